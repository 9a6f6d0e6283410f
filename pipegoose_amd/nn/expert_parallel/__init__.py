from pipegoose_amd.nn.expert_parallel.expert_context import ExpertContext
from pipegoose_amd.nn.expert_parallel.expert_parallel import ExpertParallel
from pipegoose_amd.nn.expert_parallel.layers import ExpertLayer
from pipegoose_amd.nn.expert_parallel.loss import ExpertLoss
from pipegoose_amd.nn.expert_parallel.routers import (
    SwitchNoisePolicy,
    Top1Router,
    Top2Router,
)

__all__ = [
    "ExpertParallel",
    "ExpertLayer",
    "ExpertLoss",
    "ExpertContext",
    "SwitchNoisePolicy",
    "Top1Router",
    "Top2Router",
]
