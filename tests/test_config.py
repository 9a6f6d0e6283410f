"""RuntimeConfig: defaults < yaml < env precedence + env export."""
import os

import pytest

from pipegoose_amd.config import RuntimeConfig


def test_defaults_and_env(monkeypatch):
    monkeypatch.setenv("PG_BUCKET_MB", "40")
    monkeypatch.setenv("PG_ATTN_V2", "0")
    cfg = RuntimeConfig.from_env()
    assert cfg.bucket_size_mb == 40
    assert cfg.attn_v2 == 0
    assert cfg.optimizer == "hip"


def test_yaml_env_precedence(tmp_path, monkeypatch):
    f = tmp_path / "job.yaml"
    f.write_text("bucket_size_mb: 50\nattn_v2: 0\noptimizer: foreach\n")
    monkeypatch.setenv("PG_BUCKET_MB", "60")  # env beats the file
    cfg = RuntimeConfig.from_yaml(str(f))
    assert cfg.bucket_size_mb == 60
    assert cfg.attn_v2 == 0          # file beats default
    assert cfg.optimizer == "foreach"


def test_unknown_key_rejected(tmp_path):
    f = tmp_path / "bad.yaml"
    f.write_text("not_a_knob: 1\n")
    with pytest.raises(ValueError, match="unknown config keys"):
        RuntimeConfig.from_yaml(str(f))


def test_apply_env_roundtrip(monkeypatch):
    cfg = RuntimeConfig()
    cfg.moe_grouped = 1
    cfg.apply_env()
    assert os.environ["PG_MOE_GROUPED"] == "1"
    assert RuntimeConfig.from_env().moe_grouped == 1
    monkeypatch.delenv("PG_MOE_GROUPED", raising=False)
