"""Pydantic config schema + JSON persistence.

Capability parity with the reference's pmodels.py:4-46 (Worker_Model,
Benchmark_Payload, ConfigModel) and world.py:616-722 (load/save/migration).
A "worker" here is a GPU rank on this node, not a remote webui instance, so
the identity fields are a device ordinal instead of address/port/auth.
"""
from __future__ import annotations

import json
import os
import tempfile
from typing import Dict, List, Optional

from pydantic import BaseModel, Field

CONFIG_VERSION = 1

# Canonical benchmark payload — same methodology as the reference
# (shared.py:63-77): a fixed prompt at 512x512, 20 steps, batch 1, Euler a,
# 2 warmup + 3 timed samples averaged into images-per-minute.
BENCHMARK_WARMUP_SAMPLES = 2
BENCHMARK_TIMED_SAMPLES = 3


class BenchmarkPayload(BaseModel):
    prompt: str = "A herd of cows grazing at the bottom of a sunny valley"
    negative_prompt: str = ""
    steps: int = 20
    width: int = 512
    height: int = 512
    batch_size: int = 1
    sampler_name: str = "Euler a"


class WorkerModel(BaseModel):
    """Persisted per-rank record (ref pmodels.py:12-34)."""

    label: str
    device: int = 0
    avg_ipm: float = 0.0  # canonical-payload images/minute, 0 = unbenchmarked
    eta_percent_error: List[float] = Field(default_factory=list)
    last_mpe: Optional[float] = None
    state: str = "IDLE"
    disabled: bool = False
    pixel_cap: int = 0  # max pixels (batch*W*H) per job, 0 = uncapped
    model_override: Optional[str] = None


class SettingsModel(BaseModel):
    """Engine settings (ref ui.py:363-391 Settings tab semantics)."""

    job_timeout: float = 3.0  # seconds of predicted stall before a rank is deferred
    complement_production: bool = True  # slow ranks produce bonus images
    step_scaling: bool = False  # scale complementary jobs' steps down to fit
    thin_client: bool = False  # rank 0 only orchestrates, takes no shard
    # per-task distribution toggles (ref CHANGELOG 2.3.0: separate txt2img /
    # img2img enable states); disabled -> the whole batch runs on rank 0
    distribute_txt2img: bool = True
    distribute_img2img: bool = True


class ConfigModel(BaseModel):
    version: int = CONFIG_VERSION
    workers: List[WorkerModel] = Field(default_factory=list)
    benchmark_payload: BenchmarkPayload = Field(default_factory=BenchmarkPayload)
    settings: SettingsModel = Field(default_factory=SettingsModel)
    extras: Dict[str, str] = Field(default_factory=dict)


def default_config_path() -> str:
    return os.environ.get("SDWD_CONFIG", "distributed-config.json")


def load_config(path: Optional[str] = None) -> ConfigModel:
    """Load config; tolerate a missing/corrupt file or an older schema."""
    path = path or default_config_path()
    if not os.path.exists(path):
        return ConfigModel()
    try:
        with open(path, "r", encoding="utf-8") as fh:
            raw = json.load(fh)
    except (OSError, json.JSONDecodeError):
        return ConfigModel()
    raw = _migrate(raw)
    try:
        return ConfigModel.model_validate(raw)
    except Exception:
        return ConfigModel()


def save_config(cfg: ConfigModel, path: Optional[str] = None) -> str:
    """Atomic JSON write (temp file + rename)."""
    path = path or default_config_path()
    directory = os.path.dirname(os.path.abspath(path))
    os.makedirs(directory, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=directory, prefix=".sdwd_cfg_")
    try:
        with os.fdopen(fd, "w", encoding="utf-8") as fh:
            json.dump(cfg.model_dump(), fh, indent=2)
        os.replace(tmp, path)
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)
    return path


def _migrate(raw: dict) -> dict:
    """Schema migration hook (ref world.py:632-649 legacy workers.json)."""
    version = raw.get("version", 0)
    if version == 0 and "workers" in raw and isinstance(raw["workers"], dict):
        # v0 stored workers as {label: {...}}; flatten to a list.
        raw["workers"] = [
            {"label": label, **(body or {})} for label, body in raw["workers"].items()
        ]
        raw["version"] = CONFIG_VERSION
    return raw
