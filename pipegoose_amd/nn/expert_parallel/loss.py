"""ExpertLoss: wraps a user loss fn, adding scaled router aux/z losses
(reference: nn/expert_parallel/loss.py)."""
from typing import Callable

from pipegoose_amd.nn.expert_parallel.expert_context import ExpertContext


class ExpertLoss:
    def __init__(self, loss_func: Callable, aux_weight: float = 0.01, z_weight: float = 0.1):
        self.loss_func = loss_func
        self.aux_weight = aux_weight
        self.z_weight = z_weight

    @property
    def aux_loss(self):
        return ExpertContext.get_instance().aux_losses

    @property
    def z_loss(self):
        return ExpertContext.get_instance().z_losses

    def __call__(self, *args, **kwargs):
        loss = self.loss_func(*args, **kwargs)
        ctx = ExpertContext.get_instance()
        for aux in ctx.pop_all_aux_loss():
            loss = loss + self.aux_weight * aux
        for z in ctx.pop_all_z_loss():
            loss = loss + self.z_weight * z
        return loss
