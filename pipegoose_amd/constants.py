"""Framework-wide constants."""

SEED = 69

# Checkpoint shard naming (reference-compatible: pipegoose constants.py:4-5).
CHECKPOINT_WEIGHTS_NAME = "pytorch_model_tp_{}_pp_{}.bin"
CHECKPOINT_OPTIM_NAME = "optim_states_tp_{}_pp_{}_dp_{}.bin"

# Gradient-bucket flat-buffer size for DP all-reduce.  Sized for xGMI: each
# GPU drives 7 point-to-point links at ~153 GB/s, so a ring all-reduce step
# moves bucket/N per link; 50 MB keeps per-launch latency ~100us while still
# amortizing RCCL launch overhead, and 288 GB HBM3E makes the staging buffers
# cheap.  (Reference used 25 MB, constants.py:8.)
BUCKET_SIZE_MB = 50
