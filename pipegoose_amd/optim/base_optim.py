"""Distributed optimizer protocol (reference: optim/base_optim.py)."""
from abc import ABC, abstractmethod


class BaseDistributedOptimizer(ABC):
    @property
    def defaults(self):
        return self.optim.defaults

    @property
    def param_groups(self):
        return self.optim.param_groups

    @abstractmethod
    def add_param_group(self, *args, **kwargs):
        ...

    @abstractmethod
    def load_state_dict(self, *args, **kwargs):
        ...

    @abstractmethod
    def state_dict(self, *args, **kwargs):
        ...

    @abstractmethod
    def step(self, *args, **kwargs):
        ...

    @abstractmethod
    def zero_grad(self, *args, **kwargs):
        ...
