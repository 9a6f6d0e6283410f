from pipegoose_amd.partitioning.profile import ProfileByMemory

__all__ = ["ProfileByMemory"]
