"""Pipeline P2P transport over RCCL/gloo.

Replaces the reference's TensorPipe RPC data plane (nn/pipeline_parallel/
_comm.py:9-41 pushed Packages via rpc_sync into a global queue).  Here
activations/grads move as raw tensors over torch.distributed P2P — RCCL
send/recv rides xGMI directly; shapes are negotiated once per engine run
(same [B/m, S, H] every microbatch), so steady-state transfers are a single
payload message with no metadata round trips.
"""
import torch
import torch.distributed as dist

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


class PipelineP2P:
    def __init__(self, parallel_context: ParallelContext):
        self.pc = parallel_context
        self.group = parallel_context.get_group(ParallelMode.PIPELINE)
        self._shape_cache = {}
        # Sends ride a dedicated HIP stream: an RCCL send kernel SPINS until
        # the peer's recv kernel is resident, so a recv queued behind an
        # unmatched send on the same stream deadlocks (classic 1F1B cycle:
        # rank i [send_grad -> recv_act] vs rank i-1 [send_act -> recv_grad]).
        # A separate send stream breaks every such cycle.
        self._send_streams = {}  # per-channel, created lazily
        self._recv_streams = {}

    def _device(self):
        if dist.get_backend(self.group) == "nccl":
            return self.pc.device
        return torch.device("cpu")

    # Fixed-shape fast path: caller guarantees shape/dtype via negotiate()
    def send_activation(self, tensor: torch.Tensor, dst_global: int,
                        tag: int = 0, channel: int = 0):
        payload = tensor.detach().contiguous().to(self._device())
        stream = self._send_streams.get(channel)
        if stream is None and torch.cuda.is_available():
            stream = self._send_streams[channel] = torch.cuda.Stream()
        if stream is not None and payload.is_cuda:
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                work = dist.isend(payload, dst=dst_global, group=self.group,
                                  tag=tag)
        else:
            work = dist.isend(payload, dst=dst_global, group=self.group, tag=tag)
        return work, payload  # keep payload alive until work completes

    def recv_activation(self, shape, dtype, src_global: int, tag: int = 0,
                        requires_grad: bool = False) -> torch.Tensor:
        buf = torch.empty(shape, dtype=dtype, device=self._device())
        dist.recv(buf, src=src_global, group=self.group, tag=tag)
        buf.requires_grad_(requires_grad)
        return buf

    def recv_activation_async(self, shape, dtype, src_global: int,
                              tag: int = 0, channel: int = 0):
        """Post an irecv on a dedicated per-CHANNEL recv stream (an RCCL recv
        kernel spins until matched — posted on the compute stream it would
        block everything queued behind it, and forward/backward prefetches
        must not queue behind each other either: in 1F1B a grad can be
        needed before later activations have even been sent).  Returns
        (work, buf); consume with ``work.wait()`` from the compute stream
        (inserts a stream dependency, not a host stall)."""
        stream = self._recv_streams.get(channel)
        if stream is None and torch.cuda.is_available():
            stream = self._recv_streams[channel] = torch.cuda.Stream()
        buf = torch.empty(shape, dtype=dtype, device=self._device())
        if stream is not None and buf.is_cuda:
            with torch.cuda.stream(stream):
                work = dist.irecv(buf, src=src_global, group=self.group,
                                  tag=tag)
        else:
            work = dist.irecv(buf, src=src_global, group=self.group, tag=tag)
        return work, buf
