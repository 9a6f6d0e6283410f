"""CPU unit tests for the hipGraph-decode primitives against the
established oracles (the integration parity tests cover them end-to-end;
these pin the per-op contracts)."""
import torch

from pipegoose_amd.ops.rope import _rope_ref, rope_at_position


def test_rope_at_position_matches_offset_oracle():
    torch.manual_seed(0)
    x = torch.randn(2, 4, 1, 16)
    for pos in (0, 1, 7, 100):
        ref = _rope_ref(x, 10000.0, pos_offset=pos)
        out = rope_at_position(x, 10000.0, torch.tensor([pos]))
        assert torch.allclose(out, ref, atol=1e-6), pos


def _run_bias_oracle(rank, world_size, port):
    from pipegoose_amd.models.bloom import BloomAttention, bloom_tiny
    from pipegoose_amd.testing.utils import init_parallel_context

    ctx = init_parallel_context(rank, world_size, port)
    attn = BloomAttention(bloom_tiny(), ctx)
    k_len, pos = 32, 9
    g = attn._alibi_bias_graph(k_len, torch.tensor([pos]), torch.float32)
    # oracle: the rect decode bias for q_len=1 over the filled prefix
    r = attn._alibi_bias_rect(1, pos + 1, torch.device("cpu"), torch.float32)
    assert torch.allclose(g[..., :pos + 1], r, atol=1e-6)
    assert torch.isinf(g[..., pos + 1:]).all() and (g[..., pos + 1:] < 0).all()
    ctx.destroy()


def test_alibi_graph_bias_matches_rect_oracle():
    from pipegoose_amd.testing.utils import spawn
    spawn(_run_bias_oracle, world_size=1)
