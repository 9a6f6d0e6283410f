"""Runtime configuration: one typed object for every tunable knob.

SURVEY §5 'Config / flag system': the reference had only a constants
module; here every knob the framework reads — bucket sizing, kernel
variant switches, watchdog timeouts — lives in one dataclass that loads
from defaults < YAML file < environment (PG_* variables), so a job can be
reproduced from its config dump.

Usage::

    from pipegoose_amd.config import RuntimeConfig, get_config
    cfg = get_config()                  # process-wide, env-applied
    cfg = RuntimeConfig.from_yaml("job.yaml")   # explicit file
    cfg.apply_env()                     # export back as PG_* for kernels

The HIP extension reads its knobs via getenv at first use, so apply_env()
must run before the first kernel launch to take effect there.
"""
import dataclasses
import os
from dataclasses import dataclass
from typing import Optional

_ENV_MAP = {
    "bucket_size_mb": "PG_BUCKET_MB",
    "attn_v2": "PG_ATTN_V2",
    "attn_bwd_v2": "PG_ATTN_BWD_V2",
    "attn_w4": "PG_ATTN_W4",
    "moe_grouped": "PG_MOE_GROUPED",
    "hand_gemm": "PG_HAND_GEMM",
    "bgelu_rows": "PG_BGELU_ROWS",
    "fp8_mlp": "PG_FP8_MLP",
    "fp8_attn": "PG_FP8_ATTN",
    "optimizer": "PG_OPT",
    "bench_step_timeout_s": "PG_BENCH_STEP_TIMEOUT_S",
    "bench_deadline_s": "PG_BENCH_DEADLINE_S",
    "disable_ext": "PIPEGOOSE_DISABLE_EXT",
}


@dataclass
class RuntimeConfig:
    # communication
    bucket_size_mb: int = 25          # DP/ZeRO grad bucket size (xGMI-tuned)
    # kernel variant switches (read by the HIP extension via getenv)
    attn_v2: int = 1                  # 0 forces the v1 forward
    attn_bwd_v2: int = 1              # 0 forces the v1 backward
    attn_w4: int = 0                  # 1 forces 4-wave attention WGs (A/B)
    moe_grouped: int = 0              # 1 routes mask-MoE through the bank
    hand_gemm: int = 0                # 1 forces the hand MFMA GEMM
    bgelu_rows: int = 1024            # bias-gelu bwd grid rows
    disable_ext: int = 0              # 1 disables the HIP extension
    # precision (EXPERIMENTAL: headline benchmarks stay bf16)
    fp8_mlp: int = 0                  # 1 = MLP GEMMs in OCP fp8 (ops/fp8.py)
    fp8_attn: int = 0                 # 1 = attention projections in fp8 too
    # training-loop
    optimizer: str = "hip"            # hip | fused | foreach (bench.py)
    bench_step_timeout_s: float = 300.0
    bench_deadline_s: Optional[float] = None
    seed: int = 69

    @classmethod
    def from_env(cls) -> "RuntimeConfig":
        cfg = cls()
        for field, env in _ENV_MAP.items():
            val = os.environ.get(env)
            if val is None:
                continue
            t = type(getattr(cfg, field)) if getattr(cfg, field) is not None \
                else float
            try:
                setattr(cfg, field, t(val) if t is not str else val)
            except (TypeError, ValueError):
                raise ValueError(f"bad value for {env}: {val!r}")
        return cfg

    @classmethod
    def from_yaml(cls, path: str) -> "RuntimeConfig":
        import yaml
        with open(path) as f:
            data = yaml.safe_load(f) or {}
        cfg = cls.from_env()  # env still wins over defaults, file in between
        file_cfg = cls()
        unknown = set(data) - {f.name for f in dataclasses.fields(cls)}
        if unknown:
            raise ValueError(f"unknown config keys: {sorted(unknown)}")
        for k, v in data.items():
            setattr(file_cfg, k, v)
        # precedence: defaults < yaml < env
        for f in dataclasses.fields(cls):
            env = _ENV_MAP.get(f.name)
            if env is not None and os.environ.get(env) is not None:
                continue  # env already applied via from_env
            setattr(cfg, f.name, getattr(file_cfg, f.name))
        return cfg

    def apply_env(self):
        """Export kernel-facing knobs as PG_* env vars (the HIP extension
        reads them at first use)."""
        for field, env in _ENV_MAP.items():
            val = getattr(self, field)
            if val is not None:
                os.environ[env] = str(val)
        return self

    def dump(self) -> dict:
        return dataclasses.asdict(self)


_CONFIG: Optional[RuntimeConfig] = None


def get_config() -> RuntimeConfig:
    global _CONFIG
    if _CONFIG is None:
        _CONFIG = RuntimeConfig.from_env()
    return _CONFIG
