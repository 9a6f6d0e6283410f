"""ZeRO-1 DistributedOptimizer.

Reference parity: optim/zero/optim.py:14-75 (greedy numel sharding over DP,
local step on own shard, param re-sync).  MI355X redesign of the sync:

- every rank launches async broadcasts of the owner shards all in flight
  together over xGMI (the reference did dp_size sequential broadcasts,
  optim.py:62-66);
- each owner's parameters are RE-POINTED into one persistent flat buffer
  per (owner, dtype) at setup, so the broadcast sends the live storage
  directly — no per-step flatten copy and no unflatten copy-back at all
  (the round-1 version re-flattened ~all params every step: for bloom-7b1
  that was a full extra parameter copy per step per rank).
"""
from typing import List

import torch
import torch.distributed as dist

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.optim.base_optim import BaseDistributedOptimizer
from pipegoose_amd.optim.sharding import OptimizerStateSharding


class DistributedOptimizer(BaseDistributedOptimizer):
    def __init__(self, optim: torch.optim.Optimizer,
                 parallel_context: ParallelContext,
                 parallel_mode: ParallelMode = ParallelMode.DATA,
                 grad_reduce: str = "replicate"):
        """grad_reduce="shard": gradients are REDUCED to their shard owner
        during backward (DataParallel's buckets route on the ``_zero_owner``
        tag this sets) instead of all-reduced everywhere — half the xGMI
        traffic; non-owned grads are then garbage, so use
        ``clip_grad_norm_`` below instead of a local clip."""
        assert grad_reduce in ("replicate", "shard")
        self.optim = optim
        self.parallel_context = parallel_context
        self.parallel_mode = parallel_mode
        self.grad_reduce = grad_reduce
        self._master_params = optim.param_groups  # full, for zero_grad
        self._setup_local_optim()

    def _setup_local_optim(self):
        pc = self.parallel_context
        world = pc.get_world_size(self.parallel_mode)
        if world == 1:
            self._rank_params = None
            return
        sharder = OptimizerStateSharding(self.optim.param_groups, pc, self.parallel_mode)
        partitions = sharder.shard()
        rank = pc.get_local_rank(self.parallel_mode)
        # params owned per rank (flattened order), for the post-step sync
        self._rank_params: List[List[torch.Tensor]] = [
            [p for g in partitions[r] for p in g["params"]] for r in range(world)
        ]
        if self.grad_reduce == "shard":
            for owner, params in enumerate(self._rank_params):
                for p in params:
                    if not getattr(p, "is_expert", False):
                        p._zero_owner = owner
        self.optim.param_groups = []
        for group in partitions[rank]:
            self.optim.add_param_group(group)
        self._flat_shards = None
        self._build_flat_shards()

    def _build_flat_shards(self):
        """Re-point every param's storage into one flat buffer per
        (owner, dtype): the post-step broadcast then ships the live
        storage, with no flatten/unflatten copies."""
        self._flat_shards = []
        for params in self._rank_params:
            by_dtype = {}
            for p in params:
                by_dtype.setdefault((p.dtype, p.device), []).append(p)
            flats = []
            for (dt, dev), ps in by_dtype.items():
                total = sum(p.numel() for p in ps)
                flat = torch.empty(total, dtype=dt, device=dev)
                off = 0
                for p in ps:
                    n = p.numel()
                    flat[off:off + n].copy_(p.data.reshape(-1))
                    p.data = flat[off:off + n].view_as(p.data)
                    off += n
                flats.append((flat, ps))
            self._flat_shards.append(flats)

    def _flat_valid(self) -> bool:
        for flats in self._flat_shards:
            for flat, ps in flats:
                for p in ps:
                    if (p.data.untyped_storage().data_ptr()
                            != flat.untyped_storage().data_ptr()):
                        return False  # model was moved/re-allocated
        return True

    # ------------------------------------------------------------------- api

    def add_param_group(self, *args, **kwargs):
        self.optim.add_param_group(*args, **kwargs)

    def state_dict(self, *args, **kwargs):
        return self.optim.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        self.optim.load_state_dict(*args, **kwargs)

    def step(self, *args, **kwargs):
        self.optim.step(*args, **kwargs)
        if self._rank_params is None:
            return
        pc = self.parallel_context
        group = pc.get_group(self.parallel_mode)
        ranks = pc.get_ranks_in_group(self.parallel_mode)
        if not self._flat_valid():
            # the model was moved (.to()) after setup: param storages were
            # re-allocated off the flat buffers — rebuild them once
            self._build_flat_shards()
        works = []
        for owner, flats in enumerate(self._flat_shards):
            for flat, _ps in flats:
                works.append(dist.broadcast(flat, src=ranks[owner],
                                            group=group, async_op=True))
        for w in works:
            w.wait()
        # nothing to copy back: every param IS a view of a broadcast buffer

    @torch.no_grad()
    def clip_grad_norm_(self, max_norm: float) -> torch.Tensor:
        """Global grad-norm clip under shard-reduced gradients: each rank
        norms ITS shard, the squared norms all-reduce, every rank scales the
        grads it owns (the only valid ones)."""
        pc = self.parallel_context
        world = pc.get_world_size(self.parallel_mode)
        rank = pc.get_local_rank(self.parallel_mode) if world > 1 else 0
        own = self._rank_params[rank] if self._rank_params is not None \
            else [p for g in self._master_params for p in g["params"]]
        sq = torch.zeros((), dtype=torch.float32)
        dev = None
        for p in own:
            if p.grad is not None:
                dev = p.grad.device
                sq = sq.to(dev) + p.grad.float().pow(2).sum()
        if dev is None:
            dev = torch.device("cpu")
            sq = sq.to(dev)
        if world > 1:
            # shards partition the params, so the shard-norms always combine
            # to the global norm (both grad_reduce modes)
            import torch.distributed as dist
            comm = sq if dist.get_backend() != "gloo" else sq.cpu()
            dist.all_reduce(comm, group=pc.get_group(self.parallel_mode))
            sq = comm.to(dev)
        total = sq.sqrt()
        scale = max_norm / (total + 1e-6)
        if scale < 1.0:
            for p in own:
                if p.grad is not None:
                    p.grad.mul_(scale)
        return total

    def zero_grad(self, set_to_none: bool = True):
        """Zero grads of ALL model params (not just this rank's shard)."""
        for group in self._master_params:
            for p in group["params"]:
                if p.grad is not None:
                    if set_to_none:
                        p.grad = None
                    else:
                        p.grad.zero_()
