"""Child-process entry for a server instance forked by the launcher.

This is where the persistent launcher's pre-imported modules pay off: the
fork inherits an interpreter with torch-ROCm, fma_amd and the HIP extension
already imported, so a new instance skips all Python/runtime cold-start
(the reference gets the same effect by pre-importing vLLM in the launcher
parent — reference inference_server/launcher/launcher.py:39-42, 836-885).

Isolation steps before serving (same hazards as the reference's
vllm_kickoff, reference launcher.py:840-867):
- new process group, so force-stop can killpg the whole TP tree without
  touching the launcher;
- close inherited *listening/socket* fds (a forked copy of the launcher's
  server socket would wedge launcher connections on restart); the exit
  sentinel pipe is NOT a socket and survives;
- stdout/stderr redirected to the per-instance log file.
"""

from __future__ import annotations

import os
import stat
import sys


def _close_inherited_sockets() -> None:
    try:
        fds = os.listdir("/proc/self/fd")
    except FileNotFoundError:  # pragma: no cover - non-Linux
        return
    for fd_name in fds:
        try:
            fd = int(fd_name)
            if fd <= 2:
                continue
            st = os.fstat(fd)
            if stat.S_ISSOCK(st.st_mode):
                os.close(fd)
        except (OSError, ValueError):
            continue


def kickoff(options: str, env_vars: dict, log_path: str) -> None:
    """Run the serving HTTP runtime with the given option string."""
    os.setpgrp()
    _close_inherited_sockets()
    for k, v in (env_vars or {}).items():
        os.environ[k] = str(v)
    log_fd = os.open(log_path, os.O_WRONLY | os.O_CREAT | os.O_APPEND, 0o644)
    os.dup2(log_fd, sys.stdout.fileno())
    os.dup2(log_fd, sys.stderr.fileno())
    print(f"[kickoff] pid={os.getpid()} options={options!r}", flush=True)
    from fma_amd.runtime import server
    server.main(options.split())
