"""Memory helpers (reference parity: pipegoose/utils/memory.py:4-6)."""
import torch


def get_tensor_storage_mem_loc(tensor: torch.Tensor) -> int:
    """Address of the underlying storage — used by tests to assert bucketing
    re-points grads into a flat buffer rather than copying."""
    return tensor.untyped_storage().data_ptr()


def device_memory_summary(device=None) -> dict:
    """Current/peak HIP allocator stats for the observability hooks."""
    if not torch.cuda.is_available():
        return {"allocated": 0, "peak": 0, "reserved": 0}
    return {
        "allocated": torch.cuda.memory_allocated(device),
        "peak": torch.cuda.max_memory_allocated(device),
        "reserved": torch.cuda.memory_reserved(device),
    }
