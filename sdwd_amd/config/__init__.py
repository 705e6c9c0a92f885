from .models import (
    BENCHMARK_TIMED_SAMPLES,
    BENCHMARK_WARMUP_SAMPLES,
    BenchmarkPayload,
    ConfigModel,
    SettingsModel,
    WorkerModel,
    default_config_path,
    load_config,
    save_config,
)
from .flags import add_flags, export_env

__all__ = [
    "BENCHMARK_TIMED_SAMPLES",
    "BENCHMARK_WARMUP_SAMPLES",
    "BenchmarkPayload",
    "ConfigModel",
    "SettingsModel",
    "WorkerModel",
    "default_config_path",
    "load_config",
    "save_config",
    "add_flags",
    "export_env",
]
