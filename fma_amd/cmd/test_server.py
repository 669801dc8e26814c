"""test-server: a fake inference server for control-plane e2e.

Equivalent of the reference's cmd/test-server (main.go:35-104): delayed
/health (default 47 s, the stand-in for engine cold start), atomic-bool
/sleep / /wake_up / /is_sleeping — lets the controllers be exercised with
no GPU and no model.
"""

from __future__ import annotations

import argparse
import os
import threading
import time

from fastapi import FastAPI, Query
from fastapi.responses import JSONResponse


def create_app(startup_delay: float = 47.0, start_sleeping: bool = False
               ) -> FastAPI:
    app = FastAPI(title="fma-amd test-server")
    started_at = time.time()
    sleeping = threading.Event()
    if start_sleeping:
        sleeping.set()

    @app.get("/health")
    def health():
        if time.time() - started_at < startup_delay:
            return JSONResponse({"status": "starting"}, status_code=503)
        return {"status": "OK"}

    @app.get("/is_sleeping")
    def is_sleeping():
        return {"is_sleeping": sleeping.is_set()}

    @app.post("/sleep")
    def sleep(level: int = Query(default=1)):
        sleeping.set()
        return {"status": "ok", "level": level}

    @app.post("/wake_up")
    def wake_up():
        sleeping.clear()
        return {"status": "ok"}

    @app.post("/v1/completions")
    def completions(body: dict):
        if sleeping.is_set():
            return JSONResponse({"error": "sleeping"}, status_code=409)
        return {"choices": [{"index": 0, "text": "test"}]}

    return app


def main() -> None:
    import uvicorn

    ap = argparse.ArgumentParser("fma-test-server")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--startup-delay", type=float, default=47.0)
    ap.add_argument("--start-sleeping", action="store_true")
    args = ap.parse_args()
    host = os.environ.get("FMA_BIND_HOST", "0.0.0.0")
    uvicorn.run(create_app(args.startup_delay, args.start_sleeping),
                host=host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
