"""CLI flag registration (ref preload.py:6-38).

The reference registered five ``--distributed-*`` argparse flags with the
webui host parser; here they attach to any argparse parser (bench.py, the
API server, tools) and mirror onto SDWD_* environment variables so child
ranks spawned by torchrun inherit them.
"""
from __future__ import annotations

import argparse
import os


def add_flags(parser: argparse.ArgumentParser) -> argparse.ArgumentParser:
    group = parser.add_argument_group("sdwd_amd")
    group.add_argument(
        "--sdwd-config",
        type=str,
        default=os.environ.get("SDWD_CONFIG", "distributed-config.json"),
        help="path of the persistent JSON config",
    )
    group.add_argument(
        "--sdwd-debug",
        action="store_true",
        default=os.environ.get("SDWD_DEBUG", "0") not in ("", "0", "false"),
        help="enable DEBUG logging",
    )
    group.add_argument(
        "--sdwd-devices",
        type=str,
        default=os.environ.get("SDWD_DEVICES", ""),
        help="comma-separated GPU ordinals to use (default: all visible)",
    )
    group.add_argument(
        "--sdwd-autosave",
        action="store_true",
        default=os.environ.get("SDWD_AUTOSAVE", "1") not in ("", "0", "false"),
        help="save config after runs/benchmarks",
    )
    group.add_argument(
        "--sdwd-log-file",
        type=str,
        default=os.environ.get("SDWD_LOG_FILE", ""),
        help="rotating log file path (empty = console only)",
    )
    return parser


def export_env(args: argparse.Namespace) -> None:
    """Propagate parsed flags to the environment for spawned ranks."""
    os.environ["SDWD_CONFIG"] = args.sdwd_config
    os.environ["SDWD_DEBUG"] = "1" if args.sdwd_debug else "0"
    if args.sdwd_devices:
        os.environ["SDWD_DEVICES"] = args.sdwd_devices
    os.environ["SDWD_AUTOSAVE"] = "1" if args.sdwd_autosave else "0"
    if args.sdwd_log_file:
        os.environ["SDWD_LOG_FILE"] = args.sdwd_log_file
