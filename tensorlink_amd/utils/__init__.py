from tensorlink_amd.utils.memory import MemoryEstimate, estimate_memory, get_gpu_memory  # noqa: F401
