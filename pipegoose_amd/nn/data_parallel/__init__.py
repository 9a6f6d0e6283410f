from pipegoose_amd.nn.data_parallel.data_parallel import DataParallel

__all__ = ["DataParallel"]
