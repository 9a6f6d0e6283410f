#!/usr/bin/env python3
"""Flagship training benchmark: BLOOM tokens/sec on MI355X.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`.
For N>1 the driver launches via torch.distributed.run, one rank per GPU over
RCCL; this script reads RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from the env.
Weak scaling: per-GPU microbatch fixed as N grows.

Measures BASELINE.json's metric: tokens/sec (whole node), BLOOM-560M TP2xDP2
(and BLOOM-7B1 TP2xPP2xDP2 via --model bloom-7b1, 1F1B pipeline + ZeRO-1),
bf16, synthetic data, random-init weights.
"""
import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", type=str, default="bloom-560m",
                   choices=["bloom-560m", "bloom-1b7", "bloom-7b1",
                            "bloom-tiny", "llama-7b", "llama-1b"])
    p.add_argument("--seq-len", type=int, default=2048)
    p.add_argument("--micro-batch", type=int, default=8,
                   help="per-DP-rank batch size (weak scaling)")
    p.add_argument("--tp", type=int, default=0, help="0 = auto by world size")
    p.add_argument("--pp", type=int, default=0)
    p.add_argument("--dp", type=int, default=0)
    p.add_argument("--virtual-stages", type=int, default=1,
                   help=">1 switches pp to the interleaved 1F1B schedule "
                        "(bubble (pp-1)/(v*m); experimental on RCCL)")
    p.add_argument("--microbatches", type=int, default=8,
                   help="pipeline microbatches (1F1B bubble = (pp-1)/(m+pp-1): "
                        "m=8 at pp=2 -> 11%% vs 20%% at m=4; per-rank "
                        "microbatch stays 2048 tokens)")
    p.add_argument("--cp", type=int, default=1,
                   help="context-parallel degree: sequences shard along S "
                        "over the CONTEXT group (ring attention); seq-len "
                        "is the FULL sequence")
    p.add_argument("--sp", action="store_true",
                   help="Megatron-style sequence parallelism over the TP group "
                        "(BASELINE config 5)")
    p.add_argument("--graph", action="store_true",
                   help="hipGraph-capture the training step (measured slower "
                        "on the GPU-bound default configs — opt-in)")
    p.add_argument("--moe", type=int, default=0, metavar="E",
                   help="replace MLPs with E-expert Switch MoE layers "
                        "(BASELINE config 4: bloom-1b7 with 8 experts; "
                        "all-to-all dispatch over the TP/EP group when tp>1)")
    p.add_argument("--fp8-mlp", action="store_true",
                   help="EXPERIMENTAL: run the MLP GEMMs in OCP fp8 "
                        "(dynamic e4m3/e5m2 scaling, ops/fp8.py). Never "
                        "the default: the reported dtype changes and the "
                        "headline benchmark stays bf16.")
    p.add_argument("--fp8-attn", action="store_true",
                   help="EXPERIMENTAL: also run the attention projection "
                        "GEMMs (qkv / output dense) in fp8")
    p.add_argument("--device", type=str, default=None)
    return p.parse_args()


def pick_parallelism(world_size: int, model: str, tp_arg: int, pp_arg: int, dp_arg: int):
    """BASELINE configs: bloom-560m TP2xDP2 @4; bloom-7b1 TP2xPP2xDP2 @8."""
    if tp_arg > 0 or pp_arg > 0:
        tp = max(tp_arg, 1)
        pp = max(pp_arg, 1)
        dp = dp_arg if dp_arg > 0 else world_size // (tp * pp)
        return tp, pp, dp
    if model == "bloom-7b1":
        table = {1: (1, 1, 1), 2: (2, 1, 1), 4: (2, 2, 1), 8: (2, 2, 2)}
    else:
        table = {1: (1, 1, 1), 2: (1, 1, 2), 4: (2, 1, 2), 8: (2, 1, 4)}
    return table.get(world_size, (2, 1, world_size // 2))


def main():
    args = parse_args()
    world_size = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")

    tp, pp, dp = pick_parallelism(world_size, args.model, args.tp, args.pp, args.dp)
    cp = max(1, args.cp)
    if cp > 1:
        assert world_size % (tp * pp * cp) == 0, "world != tp*pp*cp*dp"
        # cp takes its degree out of dp in the auto layout
        dp = world_size // (tp * pp * cp)

    from pipegoose_amd import ParallelContext, ParallelMode
    from pipegoose_amd.models.bloom import (
        BloomForCausalLM, bloom_1b7, bloom_560m, bloom_7b1, bloom_tiny,
        make_causal_lm_loss)
    from pipegoose_amd.nn import DataParallel
    from pipegoose_amd.nn.pipeline_parallel import PipelineParallel
    from pipegoose_amd.optim import DistributedOptimizer

    ctx = ParallelContext.from_torch(
        tensor_parallel_size=tp, pipeline_parallel_size=pp,
        data_parallel_size=dp, context_parallel_size=cp)

    use_gpu = torch.cuda.is_available()
    device = torch.device(args.device) if args.device else ctx.device
    dtype = torch.bfloat16 if use_gpu else torch.float32

    # Multi-GPU de-risking: the first real N>1 run must fail LOUDLY, never
    # hang to the harness timeout.  A per-step hang watchdog dumps every
    # thread's stack and aborts; PG_BENCH_DEADLINE_S (or a default scaled
    # to the step budget) bounds the WHOLE run the same way.
    wd = None
    hb = None
    if world_size > 1:
        from pipegoose_amd.utils.failure import HeartbeatMonitor
        from pipegoose_amd.utils.watchdog import HangWatchdog
        hb = HeartbeatMonitor(interval_s=5.0).start()
        step_timeout = float(os.environ.get("PG_BENCH_STEP_TIMEOUT_S", 300))
        # on hang: name the laggard rank(s) before dumping stacks/aborting
        wd = HangWatchdog(timeout_s=step_timeout,
                          on_hang=lambda: print(hb.report(), flush=True))
        wd.start()
        deadline = float(os.environ.get(
            "PG_BENCH_DEADLINE_S",
            step_timeout + 12.0 * (args.steps + args.warmup)))
        import faulthandler
        import threading

        def _deadline_abort():
            faulthandler.dump_traceback()
            print(f"[bench rank {rank}] deadline {deadline}s exceeded — "
                  "aborting (suspect a desynced collective)", flush=True)
            os._exit(124)

        t = threading.Timer(deadline, _deadline_abort)
        t.daemon = True
        t.start()

    cfg = {"bloom-560m": bloom_560m, "bloom-1b7": bloom_1b7,
           "bloom-7b1": bloom_7b1, "bloom-tiny": bloom_tiny,
           "llama-7b": None, "llama-1b": None}[args.model]
    is_llama = cfg is None
    if is_llama:
        from pipegoose_amd.models.llama import (LlamaForCausalLM, llama_1b,
                                                llama_7b)
        cfg = (llama_7b if args.model == "llama-7b" else llama_1b)()
        assert pp == 1 and args.moe == 0, \
            "llama bench covers tp/dp/cp (pp/moe are the bloom configs)"
    else:
        cfg = cfg()

    if args.sp:
        cfg.sequence_parallel = True
    if cp > 1:
        cfg.context_parallel = True
        assert args.seq_len % cp == 0
    torch.manual_seed(1234)
    model = LlamaForCausalLM(cfg, ctx) if is_llama else BloomForCausalLM(cfg, ctx)
    moe_loss_wrap = None
    if args.moe > 0:
        from torch import nn
        from pipegoose_amd.nn import ExpertParallel
        from pipegoose_amd.nn.expert_parallel import (ExpertLoss,
                                                      SwitchNoisePolicy,
                                                      Top1Router)
        h = cfg.hidden_size
        dense_expert = nn.Sequential(
            nn.Linear(h, 4 * h), nn.GELU(), nn.Linear(4 * h, h))
        model = ExpertParallel(
            model, args.moe, expert=dense_expert,
            router=Top1Router(SwitchNoisePolicy(), args.moe, h),
            enable_tensor_parallel=tp > 1,
            dispatch="alltoall" if tp > 1 else "mask",
            parallel_context=ctx).parallelize()
        moe_loss_wrap = ExpertLoss(lambda loss: loss)
    if pp > 1:
        v = max(1, args.virtual_stages)
        model = PipelineParallel(
            model, ctx, n_microbatches=args.microbatches,
            schedule="interleaved" if v > 1 else "1f1b", virtual_stages=v,
            loss_fn=make_causal_lm_loss(ctx)).parallelize()
    model = model.to(device=device, dtype=dtype)
    # flags OR the PG_FP8_* env knobs (config.py RuntimeConfig)
    args.fp8_mlp = args.fp8_mlp or os.environ.get("PG_FP8_MLP") == "1"
    args.fp8_attn = args.fp8_attn or os.environ.get("PG_FP8_ATTN") == "1"
    if args.fp8_mlp or args.fp8_attn:
        from pipegoose_amd.ops.fp8 import convert_linear_to_fp8
        names = []
        if args.fp8_mlp:
            names += ["dense_h_to_4h", "dense_4h_to_h", "gate_proj",
                      "up_proj", "down_proj"]
        if args.fp8_attn:
            names += ["query_key_value", ".dense", "q_proj", "k_proj",
                      "v_proj", "o_proj"]
        n_fp8 = convert_linear_to_fp8(model, names=names)
        if rank == 0:
            print(f"[bench] fp8: {n_fp8} linears converted", flush=True)
    if dp > 1:
        model = DataParallel(model, ctx).parallelize()
    if cp > 1:
        # params replicate over CP: grads all-reduce over the CONTEXT group
        model = DataParallel(model, ctx, mode=ParallelMode.CONTEXT).parallelize()

    # hipGraph capture: single-rank path only (no RCCL inside the graph);
    # kills per-kernel launch latency on the launch-bound small models.
    use_graph = (use_gpu and world_size == 1 and pp == 1 and dp == 1
                 and args.graph)
    opt_kw = dict(lr=1e-4, betas=(0.9, 0.95), capturable=use_graph)
    opt_choice = os.environ.get("PG_OPT", "hip")
    if use_gpu and not use_graph and opt_choice == "hip":
        # hand-written chunked AdamW: the whole update (p,m,v) in ONE HIP
        # kernel per step (optim/fused_adamw.py)
        from pipegoose_amd.optim.fused_adamw import FusedAdamW
        optim = FusedAdamW(model.parameters(), lr=1e-4, betas=(0.9, 0.95))
    elif use_gpu and opt_choice != "foreach":
        try:
            # torch's fused AdamW (one multi_tensor kernel; capturable)
            optim = torch.optim.AdamW(model.parameters(), fused=True, **opt_kw)
        except (RuntimeError, ValueError):
            optim = torch.optim.AdamW(model.parameters(), foreach=True, **opt_kw)
    else:
        optim = torch.optim.AdamW(model.parameters(), foreach=True, **opt_kw)
    if dp > 1:
        # grads REDUCE to their ZeRO shard owner (half the all-reduce bytes)
        optim = DistributedOptimizer(optim, ctx, grad_reduce="shard")

    B, S = args.micro_batch, args.seq_len
    # synthetic data, fixed per rank (weak scaling: per-GPU work constant)
    g = torch.Generator().manual_seed(4242 + rank)
    input_ids = torch.randint(0, cfg.vocab_size, (B, S), generator=g).to(device)
    if cp > 1:
        # every CP rank sees the SAME full sequence, keeps its S/cp shard
        g2 = torch.Generator().manual_seed(4242 + (rank // cp) * cp)
        full = torch.randint(0, cfg.vocab_size, (B, S), generator=g2)
        cpr = ctx.get_local_rank(ParallelMode.CONTEXT)
        Sl = S // cp
        input_ids = full[:, cpr * Sl:(cpr + 1) * Sl].to(device)

    def one_step(set_to_none: bool = True):
        if wd is not None:
            wd.tick()
            hb.tick()
        optim.zero_grad(set_to_none=set_to_none)
        if pp > 1:
            # the engine runs forward AND backward internally (1F1B)
            loss = model(input_ids, input_ids)
        else:
            loss = model(input_ids, labels=input_ids)
            if moe_loss_wrap is not None:
                loss = moe_loss_wrap(loss)  # + router aux/z losses
            loss.backward()
        optim.step()
        return loss

    step_fn = one_step
    if use_graph:
        # grads must be stable buffers across replays: allocate them once,
        # then capture with set_to_none=False
        for _ in range(max(args.warmup, 2)):
            one_step(set_to_none=False)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            one_step(set_to_none=False)
        step_fn = graph.replay
        torch.cuda.synchronize()
    else:
        for _ in range(args.warmup):
            one_step()

    if dist.is_initialized() and world_size > 1:
        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step_fn()
    if use_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized() and world_size > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (slowest rank defines the step time)
    if dist.is_initialized() and world_size > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    global_batch = B * dp
    tokens_per_step = global_batch * S
    tokens_per_sec = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    dtype_name = "bf16"
    if args.fp8_mlp or args.fp8_attn:
        tags = (["mlp"] if args.fp8_mlp else []) + \
               (["attn"] if args.fp8_attn else [])
        dtype_name = f"bf16+fp8-{'+'.join(tags)}(experimental)"
    if rank == 0:
        print(json.dumps({
            "metric": "tokens/sec",
            "value": round(tokens_per_sec, 1),
            "unit": "tokens/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": dtype_name if use_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "seq_len": S,
                "parallelism": f"tp{tp}pp{pp}dp{dp}" + (f"cp{cp}" if cp > 1 else "")
                               + (f"mb{args.microbatches}" if pp > 1 else "")
                               + ("sp" if args.sp else "")
                               + (f"moe{args.moe}" if args.moe else ""),
            },
        }))

    ctx.destroy()


if __name__ == "__main__":
    main()
