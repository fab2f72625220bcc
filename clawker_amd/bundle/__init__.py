from .loader import Harness, Stack, load_harness, load_stack, list_harnesses  # noqa: F401
