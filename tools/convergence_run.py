"""Real-data convergence run (reference artifact parity).

The reference's headline correctness artifact is a BLOOM-560M training run
on the imdb dataset with TP2xDP2 loss parity against the unparallelized
model (tests/convergence/run_hybrid_parallel.py:33-41).  This container has
no network, so no HF datasets: the real-text stand-in is the local Python
standard library sources (~5 MB of natural-language docstrings + code),
byte-level tokenized — real, highly structured text with a nontrivial
distribution, NOT synthetic random tokens.

Usage:
  python tools/convergence_run.py --steps 300            # 1-GPU 560M curve
  python tools/convergence_run.py --steps 200 --moe 8    # MoE (ref run_ep.py)
  python tools/convergence_run.py --model bloom-tiny --steps 40 --device cpu

Writes the loss curve as JSON lines to --out.
"""
import argparse
import glob
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def load_corpus(max_bytes=6_000_000):
    files = sorted(glob.glob("/usr/lib/python3.10/**/*.py", recursive=True))
    chunks, total = [], 0
    for f in files:
        try:
            b = open(f, "rb").read()
        except OSError:
            continue
        chunks.append(b)
        total += len(b)
        if total >= max_bytes:
            break
    data = b"\n".join(chunks)[:max_bytes]
    return torch.tensor(list(data), dtype=torch.long)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="bloom-560m",
                    choices=["bloom-560m", "bloom-tiny"])
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--seq-len", type=int, default=2048)
    ap.add_argument("--lr", type=float, default=3e-4)
    ap.add_argument("--device", default=None)
    ap.add_argument("--out", default="gpurun_out/convergence_curve.jsonl")
    ap.add_argument("--moe", type=int, default=0,
                    help="replace MLPs with E-expert Switch MoE "
                         "(mirrors the reference's run_ep.py artifact)")
    ap.add_argument("--seed", type=int, default=1234)
    ap.add_argument("--fp8-mlp", action="store_true",
                    help="run the MLP GEMMs in OCP fp8 (convergence check "
                         "for the experimental fp8 path)")
    ap.add_argument("--fp8-attn", action="store_true",
                    help="also run the attention projections in fp8")
    args = ap.parse_args()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29761")
    from pipegoose_amd import ParallelContext
    from pipegoose_amd.models.bloom import (BloomForCausalLM, bloom_560m,
                                            bloom_tiny)
    ctx = ParallelContext.from_torch()
    use_gpu = torch.cuda.is_available() and args.device != "cpu"
    device = torch.device(args.device or ("cuda" if use_gpu else "cpu"))
    dtype = torch.bfloat16 if use_gpu else torch.float32

    cfg = bloom_560m() if args.model == "bloom-560m" else bloom_tiny()
    torch.manual_seed(args.seed)
    model = BloomForCausalLM(cfg, ctx)
    loss_wrap = None
    if args.moe > 0:
        from torch import nn
        from pipegoose_amd.nn import ExpertParallel
        from pipegoose_amd.nn.expert_parallel import (ExpertLoss,
                                                      SwitchNoisePolicy,
                                                      Top1Router)
        h = cfg.hidden_size
        proto = nn.Sequential(nn.Linear(h, 4 * h), nn.GELU(),
                              nn.Linear(4 * h, h))
        model = ExpertParallel(
            model, args.moe, expert=proto,
            router=Top1Router(SwitchNoisePolicy(), args.moe, h),
            enable_tensor_parallel=False,
            parallel_context=ctx).parallelize()
        loss_wrap = ExpertLoss(lambda l: l)
    model = model.to(device=device, dtype=dtype)
    if args.fp8_mlp or args.fp8_attn:
        from pipegoose_amd.ops.fp8 import convert_linear_to_fp8
        names = (["dense_h_to_4h", "dense_4h_to_h"] if args.fp8_mlp else []) \
            + (["query_key_value", ".dense"] if args.fp8_attn else [])
        n8 = convert_linear_to_fp8(model, names=names)
        print(f"fp8: {n8} linears converted", flush=True)

    corpus = load_corpus()
    if args.model == "bloom-tiny":
        args.seq_len = min(args.seq_len, 256)
    n_win = corpus.numel() - args.seq_len - 1
    g = torch.Generator().manual_seed(args.seed)

    if use_gpu:
        from pipegoose_amd.optim.fused_adamw import FusedAdamW
        optim = FusedAdamW(model.parameters(), lr=args.lr)
    else:
        optim = torch.optim.AdamW(model.parameters(), lr=args.lr)

    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    out = open(args.out, "w")
    for step in range(args.steps):
        starts = torch.randint(0, n_win, (args.batch,), generator=g)
        ids = torch.stack([corpus[s:s + args.seq_len] for s in starts]).to(device)
        optim.zero_grad(set_to_none=True)
        loss = model(ids, labels=ids)
        if loss_wrap is not None:
            loss = loss_wrap(loss)  # + router aux/z losses
        loss.backward()
        optim.step()
        if step % 5 == 0 or step == args.steps - 1:
            rec = {"step": step, "loss": round(float(loss.detach()), 4)}
            out.write(json.dumps(rec) + "\n")
            out.flush()
            print(rec, flush=True)
    out.close()


if __name__ == "__main__":
    main()
