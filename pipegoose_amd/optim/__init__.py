from pipegoose_amd.optim.zero import DistributedOptimizer

__all__ = ["DistributedOptimizer"]
