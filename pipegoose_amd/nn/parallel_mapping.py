"""Name-pattern registry mapping HF module names to parallelization roles.

Reference parity: pipegoose/nn/parallel_mapping.py:11-37 — suffix match on the
last segments of a dotted module name.
"""
from typing import Optional, Tuple


class ParallelInfo:
    def __init__(self, *names: str, **kwargs):
        self.names = names
        self.kwargs = kwargs


class ParallelMapping:
    # subclasses set: __MAPPING__ = {"arch_name_or_*": [ParallelInfo(...), ...]}
    __MAPPING__ = {}

    @classmethod
    def _search(cls, module_name: str) -> Optional[ParallelInfo]:
        suffixes = _last_segments(module_name)
        for _, infos in cls.__MAPPING__.items():
            for info in infos:
                for pat in info.names:
                    if pat in suffixes or pat == module_name:
                        return info
        return None

    @classmethod
    def register(cls, arch: str, infos: list):
        """User-extensible: add mapping entries for a new architecture."""
        cls.__MAPPING__ = {**cls.__MAPPING__, arch: infos}


def _last_segments(name: str, n: int = 2) -> Tuple[str, ...]:
    parts = name.split(".")
    out = []
    for i in range(1, min(n, len(parts)) + 1):
        out.append(".".join(parts[-i:]))
    return tuple(out)
