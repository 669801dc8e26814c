"""One managed server instance: child process, logs, zero-poll exit watch.

Design mirrors the reference's VllmInstance semantics (reference
inference_server/launcher/launcher.py:157-340):

- fork-based spawn so the launcher's pre-imported torch-ROCm/fma_amd pay
  off in the child (launcher.py:223-228);
- UUID->index translation into HIP_VISIBLE_DEVICES (launcher.py:175-191 —
  with CUDA_VISIBLE_DEVICES in the reference);
- graceful stop: SIGTERM + join(10 s), then SIGKILL to the child's whole
  process group (launcher.py:230-258);
- zero-poll exit detection: the multiprocessing sentinel fd is registered
  on the asyncio loop; the kernel makes it readable when the child dies
  (launcher.py:260-293).
"""

from __future__ import annotations

import asyncio
import multiprocessing
import os
import signal
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from fma_amd.api import contracts
from fma_amd.launcher.gputranslator import GpuTranslator
from fma_amd.runtime.kickoff import kickoff

STATUS_RUNNING = contracts.INSTANCE_STATUS_RUNNING
STATUS_STOPPED = contracts.INSTANCE_STATUS_STOPPED


@dataclass
class ServerConfig:
    """Wire schema of an instance config (reference launcherclient.go:88-94:
    VllmConfig{options, gpu_uuids, env_vars, annotations})."""

    options: str
    gpu_uuids: List[str] = field(default_factory=list)
    env_vars: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, object]:
        return {
            "options": self.options,
            "gpu_uuids": list(self.gpu_uuids),
            "env_vars": dict(self.env_vars),
            "annotations": dict(self.annotations),
        }

    @staticmethod
    def from_dict(d: Dict[str, object]) -> "ServerConfig":
        return ServerConfig(
            options=str(d.get("options", "")),
            gpu_uuids=list(d.get("gpu_uuids") or []),
            env_vars=dict(d.get("env_vars") or {}),
            annotations=dict(d.get("annotations") or {}),
        )


class ServerInstance:
    def __init__(self, instance_id: str, config: ServerConfig,
                 translator: GpuTranslator, log_dir: str = "/tmp"):
        self.instance_id = instance_id
        self.config = config
        self.translator = translator
        self.log_path = os.path.join(
            log_dir, f"launcher-{os.getpid()}-server-{instance_id}.log")
        self.process: Optional[multiprocessing.Process] = None
        self.exit_code: Optional[int] = None
        self.created_at = time.time()
        self._stopped = False

    # -- lifecycle -----------------------------------------------------------

    def start(self) -> None:
        env = dict(self.config.env_vars)
        if self.config.gpu_uuids:
            env[contracts.VISIBLE_DEVICES_ENV] = \
                self.translator.visible_devices_value(self.config.gpu_uuids)
        open(self.log_path, "a").close()
        ctx = multiprocessing.get_context("fork")
        self.process = ctx.Process(
            target=kickoff,
            args=(self.config.options, env, self.log_path),
            name=f"fma-server-{self.instance_id}")
        self.process.start()

    @property
    def pid(self) -> Optional[int]:
        return self.process.pid if self.process else None

    @property
    def status(self) -> str:
        if self.process is None or self._stopped:
            return STATUS_STOPPED
        return STATUS_RUNNING if self.process.is_alive() else STATUS_STOPPED

    def register_sentinel(self, loop: asyncio.AbstractEventLoop,
                          on_exit: Callable[["ServerInstance", Optional[int]],
                                            None]) -> None:
        """Zero-poll exit detection via the child's sentinel fd."""
        if self.process is None:
            return
        sentinel = self.process.sentinel

        def _cb() -> None:
            try:
                loop.remove_reader(sentinel)
            except (ValueError, OSError):
                pass
            self.process.join(timeout=1)
            self.exit_code = self.process.exitcode
            on_exit(self, self.exit_code)

        loop.add_reader(sentinel, _cb)

    def stop(self, grace_seconds: float = 10.0) -> None:
        """SIGTERM, wait, then SIGKILL the whole child process group."""
        if self.process is None or self._stopped:
            self._stopped = True
            return
        pid = self.process.pid
        if self.process.is_alive() and pid:
            try:
                os.kill(pid, signal.SIGTERM)
            except ProcessLookupError:
                pass
            self.process.join(timeout=grace_seconds)
            if self.process.is_alive():
                try:
                    # the child called setpgrp, so its pgid == its pid and
                    # this reaps TP workers too
                    os.killpg(pid, signal.SIGKILL)
                except (ProcessLookupError, PermissionError):
                    try:
                        os.kill(pid, signal.SIGKILL)
                    except ProcessLookupError:
                        pass
                self.process.join(timeout=5)
        self.exit_code = self.process.exitcode
        self._stopped = True

    def delete_log(self) -> None:
        try:
            os.unlink(self.log_path)
        except FileNotFoundError:
            pass

    # -- logs ----------------------------------------------------------------

    def log_size(self) -> int:
        try:
            return os.path.getsize(self.log_path)
        except OSError:
            return 0

    def get_log_bytes(self, start: int, length: int) -> bytes:
        with open(self.log_path, "rb") as f:
            f.seek(start)
            return f.read(length)

    # -- wire ----------------------------------------------------------------

    def state_dict(self) -> Dict[str, object]:
        d = {"instance_id": self.instance_id, "status": self.status}
        if self.exit_code is not None:
            d["exit_code"] = self.exit_code
        d.update(self.config.to_dict())
        return d
