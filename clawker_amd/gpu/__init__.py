from .inventory import GPUDevice, GPUInventory  # noqa: F401
from .allocator import GPUAllocator, GPUAllocationError  # noqa: F401
