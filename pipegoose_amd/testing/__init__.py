from pipegoose_amd.testing.utils import (
    find_free_port,
    init_parallel_context,
    spawn,
)

__all__ = ["spawn", "init_parallel_context", "find_free_port"]
