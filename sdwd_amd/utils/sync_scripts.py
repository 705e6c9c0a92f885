"""User sync-script runner (ref C21, ui.py:26-55 + scripts/user/).

The reference let users drop a ``sync*`` shell script (e.g. an rclone model
sync) into scripts/user/ and run it from the Utils tab. Same contract here:
the first executable matching sync* in the user-script directory runs with
the repo root as cwd; stdout/stderr are captured into the log ring buffer.
"""
from __future__ import annotations

import glob
import os
import stat
import subprocess
from typing import Optional, Tuple

from . import get_logger

log = get_logger("sync")

DEFAULT_DIR = os.environ.get("SDWD_USER_SCRIPTS", "scripts/user")


def find_sync_script(directory: Optional[str] = None) -> Optional[str]:
    directory = directory or DEFAULT_DIR
    for cand in sorted(glob.glob(os.path.join(directory, "sync*"))):
        if os.path.isfile(cand):
            return cand
    return None


def run_sync_script(
    directory: Optional[str] = None, timeout: float = 600.0
) -> Tuple[int, str]:
    """Run the user's sync script; returns (returncode, combined output)."""
    script = find_sync_script(directory)
    if script is None:
        log.warning("no sync* script found in %s", directory or DEFAULT_DIR)
        return (127, "no sync script found")
    if not os.access(script, os.X_OK):
        os.chmod(script, os.stat(script).st_mode | stat.S_IXUSR)
    log.info("running user sync script %s", script)
    try:
        proc = subprocess.run(
            [os.path.abspath(script)],
            capture_output=True,
            text=True,
            timeout=timeout,
            cwd=os.getcwd(),
        )
        output = (proc.stdout or "") + (proc.stderr or "")
        for line in output.strip().splitlines()[-16:]:
            log.info("[sync] %s", line)
        return (proc.returncode, output)
    except subprocess.TimeoutExpired:
        log.error("sync script timed out after %.0fs", timeout)
        return (124, "timeout")
