"""Native LLaMA family: TP oracle parity, SP parity, PP partitioning.

Oracle pattern (reference tests/nn/tensor_parallel/test_tensor_parallel.py):
run the tp=1 model in one process, assert the tp=2 sharded run matches.
"""
import torch

from pipegoose_amd.models.llama import (LlamaForCausalLM, llama_tiny)
from pipegoose_amd.testing.utils import init_parallel_context, spawn


def _tiny(sp=False):
    cfg = llama_tiny()
    cfg.sequence_parallel = sp
    return cfg


def test_llama_single_rank_forward_backward():
    spawn(_run_llama_smoke, world_size=1)


def _run_llama_smoke(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(21)
    model = LlamaForCausalLM(_tiny(), ctx)
    ids = torch.randint(0, 256, (2, 12))
    logits = model(ids)
    assert logits.shape == (2, 12, 256)
    loss = model(ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)
    assert model.model.embed_tokens.weight.grad is not None
    ctx.destroy()


def _run_tp2_oracle(rank, world_size, port):
    import torch.nn.functional as TF
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=2)
    # oracle: same init with a WORLD-1-equivalent full model is impossible
    # in-process under tp=2; instead verify sharded logits gather to the
    # reference computed manually from gathered weights.
    torch.manual_seed(23)
    model = LlamaForCausalLM(_tiny(), ctx)
    torch.manual_seed(24)
    ids = torch.randint(0, 256, (2, 8))
    logits_sharded = model(ids)  # [B, S, V/2]
    assert logits_sharded.shape == (2, 8, 128)
    loss = model(ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)
    # vocab-parallel CE must equal CE over gathered logits
    import torch.distributed as dist
    from pipegoose_amd.distributed.parallel_mode import ParallelMode
    full = [torch.empty_like(logits_sharded) for _ in range(2)]
    dist.all_gather(full, logits_sharded.contiguous(),
                    group=ctx.get_group(ParallelMode.TENSOR))
    full = torch.cat(full, dim=-1)
    ref = TF.cross_entropy(
        full[:, :-1].reshape(-1, 256).float(), ids[:, 1:].reshape(-1))
    assert torch.allclose(loss, ref, atol=1e-5), (loss - ref).abs()
    ctx.destroy()


def test_llama_tp2_vocab_parallel_loss():
    spawn(_run_tp2_oracle, world_size=2)


def _run_sp_parity(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=2)
    torch.manual_seed(25)
    ref = LlamaForCausalLM(_tiny(sp=False), ctx)
    torch.manual_seed(25)
    sp = LlamaForCausalLM(_tiny(sp=True), ctx)
    torch.manual_seed(26)
    ids = torch.randint(0, 256, (2, 8))
    l_ref = ref(ids, labels=ids)
    l_sp = sp(ids, labels=ids)
    assert torch.allclose(l_ref, l_sp, atol=1e-5)
    l_ref.backward()
    l_sp.backward()
    for (n, p1), p2 in zip(ref.named_parameters(), sp.parameters()):
        if p1.grad is None:
            continue
        assert torch.allclose(p1.grad, p2.grad, atol=1e-4), n
    ctx.destroy()


def test_llama_sp_matches_tp_tp2():
    spawn(_run_sp_parity, world_size=2)


def _run_pp_partition(rank, world_size, port):
    from pipegoose_amd.nn.pipeline_parallel.partitioner import UniformPartitioner
    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(27)
    model = LlamaForCausalLM(_tiny(), ctx)
    stages = UniformPartitioner(model, ctx).split(n_partitions=2)
    assert len(stages) == 2
    n_params = sum(p.numel() for s in stages for p in s.parameters())
    assert n_params == sum(p.numel() for p in model.parameters())
    ctx.destroy()


def test_llama_pp_partition():
    spawn(_run_pp_partition, world_size=1)


def _run_generate_tp2(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=2)
    torch.manual_seed(50)
    model = LlamaForCausalLM(_tiny(), ctx)
    ids = torch.randint(0, 256, (2, 6))
    out = model.generate(ids, max_new_tokens=5)
    assert out.shape == (2, 11)
    # oracle: per-step argmax over the gathered full logits
    import torch.distributed as dist
    from pipegoose_amd.distributed.parallel_mode import ParallelMode
    cur = ids
    for _ in range(5):
        local = model(cur)[:, -1]
        shards = [torch.empty_like(local) for _ in range(2)]
        dist.all_gather(shards, local.contiguous(),
                        group=ctx.get_group(ParallelMode.TENSOR))
        nxt = torch.cat(shards, -1).float().argmax(-1)
        cur = torch.cat([cur, nxt.unsqueeze(-1)], dim=-1)
    assert torch.equal(out, cur)
    ctx.destroy()


def test_llama_generate_tp2_matches_gathered_argmax():
    spawn(_run_generate_tp2, world_size=2)


def _run_llama_kv_cache(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(61)
    model = LlamaForCausalLM(_tiny(), ctx)
    model.eval()
    ids = torch.randint(0, 256, (2, 6))
    with torch.no_grad():
        logits_c, past = model(ids, use_cache=True)
        logits_f = model(ids)
        assert torch.allclose(logits_c, logits_f, atol=1e-5)
        nxt = logits_f[:, -1].argmax(-1, keepdim=True)
        step_logits, past = model(nxt, past=past, use_cache=True)
        full = model(torch.cat([ids, nxt], dim=-1))
        assert torch.allclose(step_logits[:, -1], full[:, -1], atol=1e-5), \
            (step_logits[:, -1] - full[:, -1]).abs().max()
    ctx.destroy()


def test_llama_kv_cache_decode():
    spawn(_run_llama_kv_cache, world_size=1)


def _gqa_tiny():
    cfg = llama_tiny()
    cfg.n_kv_head = 2  # 4 q heads share 2 kv heads
    return cfg


def _run_gqa_smoke(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(90)
    model = LlamaForCausalLM(_gqa_tiny(), ctx)
    attn = model.model.layers[0].self_attn
    assert attn.num_kv_heads == 2 and attn.kv_group == 2
    assert attn.k_proj.weight.shape[0] == 2 * 16  # kv heads * head_dim
    ids = torch.randint(0, 256, (2, 12))
    loss = model(ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)
    assert attn.k_proj.weight.grad is not None
    # KV-cached decode matches full recompute
    model.zero_grad()
    model.eval()
    with torch.no_grad():
        logits, past = model(ids, use_cache=True)
        nxt = logits[:, -1].argmax(-1, keepdim=True)
        step, _ = model(nxt, past=past, use_cache=True)
        full = model(torch.cat([ids, nxt], -1))
        assert torch.allclose(step[:, -1], full[:, -1], atol=1e-5)
    ctx.destroy()


def test_llama_gqa_smoke_and_cache():
    spawn(_run_gqa_smoke, world_size=1)


def _run_gqa_tp2(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=2)
    torch.manual_seed(91)
    model = LlamaForCausalLM(_gqa_tiny(), ctx)
    assert model.model.layers[0].self_attn.num_kv_heads == 1
    ids = torch.randint(0, 256, (2, 8))
    loss = model(ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)
    ctx.destroy()


def test_llama_gqa_tp2():
    spawn(_run_gqa_tp2, world_size=2)


def _run_sample_sync_tp2(rank, world_size, port):
    """Stochastic sampling under TP must produce identical sequences on all
    ranks (rank 0 samples, the group broadcasts)."""
    import torch.distributed as dist
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=2)
    torch.manual_seed(95)
    model = LlamaForCausalLM(_tiny(), ctx)
    torch.manual_seed(1000 + rank)  # deliberately DIFFERENT RNG per rank
    ids = torch.randint(0, 256, (2, 6))
    # same prompt everywhere (re-seed for data only)
    torch.manual_seed(96)
    ids = torch.randint(0, 256, (2, 6))
    torch.manual_seed(2000 + rank)  # diverge RNG again before sampling
    out = model.generate(ids, max_new_tokens=5, temperature=0.8, top_k=5)
    peers = [torch.empty_like(out) for _ in range(2)]
    dist.all_gather(peers, out)
    assert torch.equal(peers[0], peers[1]), "sampled sequences diverged"
    ctx.destroy()


def test_sampling_synced_across_tp_ranks():
    spawn(_run_sample_sync_tp2, world_size=2)


def _run_llama_graph_decoder(rank, world_size, port):
    """Graph-mode static decode path (device-pos RoPE + full-length masked
    attention, incl. the GQA expand) must match generate()'s greedy tokens."""
    import dataclasses
    from pipegoose_amd.models.graph_decode import GraphDecoder

    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(55)
    cfg = dataclasses.replace(llama_tiny(), n_kv_head=2)  # exercise GQA
    model = LlamaForCausalLM(cfg, ctx)
    model.eval()
    torch.manual_seed(56)
    prompt = torch.randint(0, 256, (2, 9))

    ref = model.generate(prompt, max_new_tokens=7)
    dec = GraphDecoder(model, batch_size=2, max_len=32)
    out = dec.generate(prompt, max_new_tokens=7)
    assert torch.equal(out, ref[:, -7:]), (out, ref)
    ctx.destroy()


def test_llama_graph_decoder_matches_generate_cpu():
    spawn(_run_llama_graph_decoder, world_size=1)
