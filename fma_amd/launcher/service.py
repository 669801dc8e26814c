"""Launcher REST service: CRUDL + logs + watch for server instances.

API surface is wire-compatible with the reference launcher (reference
inference_server/launcher/launcher.py:534-800, docs/launcher.md:191-532) so
the dual-pods controller's client works against either:

  GET    /                              service info
  GET    /health                        {"status": "OK"}
  POST   /v2/vllm/instances             create (auto id) -> 201
  PUT    /v2/vllm/instances/{id}        create (named)   -> 201 | 409
  GET    /v2/vllm/instances             list + revision
  GET    /v2/vllm/instances/{id}        state | 404
  DELETE /v2/vllm/instances/{id}        stop+delete | 404
  DELETE /v2/vllm/instances             stop+delete all
  GET    /v2/vllm/instances/{id}/log    RFC 9110 byte ranges, 1 MiB cap
  GET    /v2/vllm/instances/watch       NDJSON watch (?since=N, 410 Gone)

The path keeps the reference's literal "/v2/vllm" segment for client
compatibility even though the engine underneath is fma_amd's own runtime.
"""

from __future__ import annotations

import asyncio
import json
import re
import uuid as uuidlib
from typing import Dict, Optional

from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse, StreamingResponse

from fma_amd.api import contracts
from fma_amd.launcher.broadcaster import EventBroadcaster, RevisionTooOld
from fma_amd.launcher.gputranslator import GpuTranslator
from fma_amd.launcher.instance import (STATUS_RUNNING, ServerConfig,
                                       ServerInstance)

LOG_BYTES_CAP = 1 << 20  # 1 MiB per request, as in the reference API


class InstanceManager:
    def __init__(self, translator: Optional[GpuTranslator] = None,
                 log_dir: str = "/tmp"):
        self.translator = translator or GpuTranslator()
        self.log_dir = log_dir
        self.instances: Dict[str, ServerInstance] = {}
        self.broadcaster = EventBroadcaster()
        self.loop: Optional[asyncio.AbstractEventLoop] = None

    def attach_loop(self, loop: asyncio.AbstractEventLoop) -> None:
        self.loop = loop
        self.broadcaster.attach_loop(loop)

    # -- operations ----------------------------------------------------------

    def create_instance(self, config: ServerConfig,
                        instance_id: Optional[str] = None) -> ServerInstance:
        iid = instance_id or str(uuidlib.uuid4())
        if iid in self.instances:
            raise KeyError(iid)
        inst = ServerInstance(iid, config, self.translator, self.log_dir)
        inst.start()
        self.instances[iid] = inst
        rev = self.broadcaster.next_revision()
        self.broadcaster.append("CREATED", iid, rev,
                                {"status": inst.status})
        if self.loop is not None:
            inst.register_sentinel(self.loop, self._on_instance_exit)
        return inst

    def _on_instance_exit(self, inst: ServerInstance,
                          exit_code: Optional[int]) -> None:
        if inst.instance_id not in self.instances:
            return  # exit caused by deletion: DELETED already tells the story
        rev = self.broadcaster.next_revision()
        self.broadcaster.append("STOPPED", inst.instance_id, rev,
                                {"exit_code": exit_code})

    def delete_instance(self, iid: str) -> ServerInstance:
        inst = self.instances.pop(iid)
        inst.stop()
        inst.delete_log()
        rev = self.broadcaster.next_revision()
        self.broadcaster.append("DELETED", iid, rev)
        return inst

    def stop_all(self) -> None:
        for iid in list(self.instances):
            self.delete_instance(iid)

    def list_state(self) -> Dict[str, object]:
        states = [inst.state_dict() for inst in self.instances.values()]
        running = sum(1 for s in states if s["status"] == STATUS_RUNNING)
        return {
            "total_instances": len(states),
            "running_instances": running,
            "instances": states,
            "revision": self.broadcaster.revision,
        }


_RANGE_RE = re.compile(r"^bytes=(\d*)-(\d*)$")


def _parse_range(header: str, size: int):
    """Returns (start, length, status) or raises ValueError(status)."""
    m = _RANGE_RE.match(header.strip())
    if not m or (not m.group(1) and not m.group(2)):
        raise ValueError(400)
    if size == 0:
        raise ValueError(416)
    if not m.group(1):  # suffix range: last N bytes
        n = int(m.group(2))
        start = max(size - n, 0)
        end = size - 1
    else:
        start = int(m.group(1))
        if start >= size:
            raise ValueError(416)
        end = int(m.group(2)) if m.group(2) else size - 1
        if end < start:
            raise ValueError(400)
    end = min(end, size - 1)
    length = min(end - start + 1, LOG_BYTES_CAP)
    return start, length


def create_app(manager: Optional[InstanceManager] = None) -> FastAPI:
    app = FastAPI(title="fma-amd launcher", version="2.0")
    mgr = manager or InstanceManager()
    app.state.manager = mgr

    @app.on_event("startup")
    async def _startup():
        mgr.attach_loop(asyncio.get_running_loop())
        # instances created before the loop existed (tests): register now
        for inst in mgr.instances.values():
            inst.register_sentinel(mgr.loop, mgr._on_instance_exit)

    @app.on_event("shutdown")
    async def _shutdown():
        mgr.stop_all()

    @app.get("/")
    def index():
        root = contracts.LAUNCHER_API_ROOT
        return {
            "name": "Multi-Instance Server Management API (fma-amd)",
            "version": "2.0",
            "endpoints": {
                "index": "GET /",
                "health": "GET /health",
                "create_instance": f"POST {root}",
                "create_named_instance": f"PUT {root}/{{instance_id}}",
                "delete_instance": f"DELETE {root}/{{instance_id}}",
                "delete_all_instances": f"DELETE {root}",
                "get_instance_status": f"GET {root}/{{instance_id}}",
                "get_all_instances": f"GET {root}",
                "get_instance_logs": f"GET {root}/{{instance_id}}/log",
                "watch_instances": f"GET {root}/watch",
            },
        }

    @app.get("/health")
    def health():
        return {"status": "OK"}

    def _created_response(inst: ServerInstance) -> JSONResponse:
        body = inst.state_dict()
        body["revision"] = mgr.broadcaster.revision
        return JSONResponse(body, status_code=201)

    @app.post(contracts.LAUNCHER_API_ROOT)
    def create_instance(config: dict):
        try:
            inst = mgr.create_instance(ServerConfig.from_dict(config))
        except Exception as e:  # pragma: no cover - spawn failure
            return JSONResponse({"error": str(e)}, status_code=500)
        return _created_response(inst)

    @app.put(contracts.LAUNCHER_API_ROOT + "/{instance_id}")
    def create_named_instance(instance_id: str, config: dict):
        if instance_id in mgr.instances:
            return JSONResponse(
                {"error": f"instance {instance_id} already exists"},
                status_code=409)
        try:
            inst = mgr.create_instance(ServerConfig.from_dict(config),
                                       instance_id)
        except KeyError:
            return JSONResponse(
                {"error": f"instance {instance_id} already exists"},
                status_code=409)
        except Exception as e:  # pragma: no cover
            return JSONResponse({"error": str(e)}, status_code=500)
        return _created_response(inst)

    @app.get(contracts.LAUNCHER_API_ROOT + "/watch")
    async def watch(request: Request, since: int = 0):
        try:
            mgr.broadcaster.check_since(since)
        except RevisionTooOld as e:
            return JSONResponse({"error": str(e)}, status_code=410)
        agen = mgr.broadcaster.watch(since)

        async def stream():
            try:
                async for ev in agen:
                    if await request.is_disconnected():
                        return
                    yield json.dumps(ev) + "\n"
            except RevisionTooOld as e:
                # Slow consumer overtaken by buffer eviction: emit one
                # terminal error line (status is already sent) and close so
                # the client re-LISTs, as with a fresh 410.
                yield json.dumps({"error": str(e), "code": 410}) + "\n"
                return
            except asyncio.CancelledError:  # pragma: no cover
                return

        return StreamingResponse(stream(), media_type="application/x-ndjson")

    @app.get(contracts.LAUNCHER_API_ROOT + "/{instance_id}")
    def get_instance(instance_id: str):
        inst = mgr.instances.get(instance_id)
        if inst is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        body = inst.state_dict()
        body["revision"] = mgr.broadcaster.revision
        return body

    @app.get(contracts.LAUNCHER_API_ROOT)
    def list_instances():
        return mgr.list_state()

    @app.delete(contracts.LAUNCHER_API_ROOT + "/{instance_id}")
    def delete_instance(instance_id: str):
        if instance_id not in mgr.instances:
            return JSONResponse({"error": "not found"}, status_code=404)
        inst = mgr.delete_instance(instance_id)
        body = inst.state_dict()
        body["status"] = "stopped"
        body["revision"] = mgr.broadcaster.revision
        return body

    @app.delete(contracts.LAUNCHER_API_ROOT)
    def delete_all():
        n = len(mgr.instances)
        mgr.stop_all()
        return {"status": "stopped", "deleted": n,
                "revision": mgr.broadcaster.revision}

    @app.get(contracts.LAUNCHER_API_ROOT + "/{instance_id}/log")
    def get_log(instance_id: str, request: Request):
        inst = mgr.instances.get(instance_id)
        if inst is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        size = inst.log_size()
        range_header = request.headers.get("range")
        if range_header is None:
            data = inst.get_log_bytes(0, min(size, LOG_BYTES_CAP))
            return Response(content=data, media_type="text/plain",
                            headers={"Content-Length": str(len(data))})
        try:
            start, length = _parse_range(range_header, size)
        except ValueError as e:
            code = e.args[0]
            if code == 416:
                return Response(status_code=416,
                                headers={"Content-Range": f"bytes */{size}"})
            return JSONResponse({"error": "malformed Range"}, status_code=400)
        data = inst.get_log_bytes(start, length)
        end = start + len(data) - 1
        return Response(
            content=data, status_code=206, media_type="text/plain",
            headers={"Content-Range": f"bytes {start}-{end}/{size}"})

    return app


def main() -> None:
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser("fma-launcher")
    ap.add_argument("--port", type=int, default=contracts.LAUNCHER_SERVICE_PORT)
    ap.add_argument("--host", default=None)
    ap.add_argument("--gpu-mode", default=None,
                    choices=[None, "real", "gpu-map", "naive"])
    ap.add_argument("--log-dir", default="/tmp")
    args = ap.parse_args()

    # Pre-import the heavy modules so forked instances skip cold-start —
    # the core launcher trick (reference launcher.py:39-42) — and pre-mmap
    # any known model checkpoints so a swapped-in server's weight load
    # hits the page cache (reference docs/dual-pods.md:599-608 analog).
    import os as _os2
    for d in (_os2.environ.get("FMA_PREMAP_MODELS") or "").split(":"):
        if d and _os2.path.isdir(d):
            from fma_amd.models.loader import premap_safetensors
            premap_safetensors(d)
    import torch  # noqa: F401
    import fma_amd.runtime.engine  # noqa: F401
    import fma_amd.runtime.server  # noqa: F401
    # CRITICAL: do NOT touch torch.cuda here — the launcher forks serving
    # instances, and a child forked from a HIP-initialized parent launches
    # kernels ~1000x slower (measured: 108 s weight init for 15 GiB).
    from fma_amd.utils.gpus import gpu_present
    if gpu_present():
        import fma_amd._C  # noqa: F401  HIP actuator must be importable

    import os as _os
    host = args.host or _os.environ.get("FMA_BIND_HOST", "0.0.0.0")
    mgr = InstanceManager(GpuTranslator(args.gpu_mode), args.log_dir)
    app = create_app(mgr)
    uvicorn.run(app, host=host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
