#!/usr/bin/env bash
# Full validation recipe (what CI / the round driver runs, collected in one
# place).  CPU parts run anywhere; GPU parts need an MI355X (run them via
# gpurun on the build host).
set -euxo pipefail
cd "$(dirname "$0")/.."

# 1. build the gfx950 extension in-tree (cross-compiles without a GPU)
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

# 2. CPU test suite (multi-process paths run on gloo/localhost)
python -m pytest tests/ -x -q -m "not gpu"

# 3. multi-rank launch smokes over the exact driver entry (torchrun, CPU)
python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29801 \
    bench.py --gpus 2 --steps 2 --warmup 1 --model bloom-tiny \
    --seq-len 128 --micro-batch 4
python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
    --master-addr 127.0.0.1 --master-port 29803 \
    bench.py --gpus 4 --steps 2 --warmup 1 --model bloom-tiny \
    --tp 2 --dp 2 --seq-len 128 --micro-batch 4
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
    --master-addr 127.0.0.1 --master-port 29802 \
    bench.py --gpus 8 --steps 2 --warmup 1 --model bloom-tiny \
    --tp 2 --pp 2 --dp 2 --seq-len 128 --micro-batch 8
python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29804 \
    bench.py --gpus 2 --steps 2 --warmup 1 --model bloom-tiny \
    --cp 2 --seq-len 128 --micro-batch 4
python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29805 \
    bench.py --gpus 2 --steps 2 --warmup 1 --model bloom-tiny \
    --tp 2 --moe 4 --seq-len 128 --micro-batch 4

# 4. GPU tier (on an MI355X box)
if python -c "import torch; raise SystemExit(0 if torch.cuda.is_available() else 1)"; then
    python -m pytest tests/ -x -q -m gpu
    python __graft_entry__.py smoke
    python bench.py --steps 10 --warmup 3
    python bench.py --model bloom-7b1 --steps 5 --warmup 2
    python tools/attn_bench.py
    python tools/decode_bench.py
    # race-detection lane (serialized kernels/copies; see the script)
    bash scripts/race_check.sh
fi
echo "ALL VALIDATION PASSED"
