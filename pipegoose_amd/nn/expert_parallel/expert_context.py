"""Process-global accumulator for router aux/z losses (reference:
nn/expert_parallel/expert_context.py)."""
from typing import List

import torch


class ExpertContext:
    _instance = None

    def __init__(self):
        self.aux_losses: List[torch.Tensor] = []
        self.z_losses: List[torch.Tensor] = []

    @classmethod
    def get_instance(cls) -> "ExpertContext":
        if cls._instance is None:
            cls._instance = ExpertContext()
        return cls._instance

    def push_aux_loss(self, loss):
        self.aux_losses.append(loss)

    def pop_all_aux_loss(self):
        losses, self.aux_losses = self.aux_losses, []
        return losses

    def push_z_loss(self, loss):
        self.z_losses.append(loss)

    def pop_all_z_loss(self):
        losses, self.z_losses = self.z_losses, []
        return losses
