from .sampler import RocmSampler, GpuSample  # noqa: F401
