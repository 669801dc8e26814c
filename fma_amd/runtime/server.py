"""Serving HTTP runtime: the process the launcher starts per instance.

Speaks the exact wire surface the dual-pods controller drives on an
inference server (reference pkg/controller/dual-pods/inference-server.go:
1497 wake, 1712 sleep, 1985 is_sleeping; SleepState JSON shape
pkg/api/interface.go:131-135), plus a minimal OpenAI-style completion
endpoint for smoke traffic:

  GET  /health           -> {"status": "OK"} once serving
  GET  /is_sleeping      -> {"is_sleeping": bool}
  POST /sleep?level=1    -> offload weights to pinned host DRAM
  POST /wake_up          -> restore weights (RCCL barrier gates readiness)
  GET  /v1/models        -> model card
  POST /v1/completions   -> {"prompt", "max_tokens"}
  GET  /stats            -> engine counters (sleep/wake seconds etc.)

CLI options mirror the reference's ModelServerConfig.Options surface
(reference api/fma/v1alpha1/inferenceserverconfig_types.go:35-62):
``--model``, ``--port``, ``--tensor-parallel-size``, ``--enable-sleep-mode``
(always on here), ``--max-model-len``, ``--seed``; unknown options are
tolerated so reference ISC manifests keep working.

TP instances: this process is rank 0; it forks ranks 1..N-1 before binding
the port (see fma_amd/parallel/tp.py).
"""

from __future__ import annotations

import argparse
import multiprocessing
import os
import socket
import threading
import time
from typing import List, Optional

from fastapi import FastAPI, Query
from fastapi.responses import JSONResponse

from fma_amd.models.llama import LlamaConfig


def parse_options(options: str) -> argparse.Namespace:
    ap = argparse.ArgumentParser(prog="fma-serve", add_help=False)
    ap.add_argument("--model", default="tiny")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--host", default=None)
    ap.add_argument("--tensor-parallel-size", type=int, default=1)
    ap.add_argument("--enable-sleep-mode", action="store_true")
    ap.add_argument("--expert-parallel", action="store_true",
                    help="MoE: partition whole experts across TP ranks "
                         "(attention stays Megatron-TP)")
    ap.add_argument("--max-model-len", type=int, default=None)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--served-model-name", default=None)
    ap.add_argument("--start-asleep", action="store_true",
                    help="immediately sleep after load (pre-warmed instance)")
    args, unknown = ap.parse_known_args(options.split())
    args.unknown = unknown
    if args.host is None:
        # node agent / launcher export the Pod's loopback identity
        args.host = os.environ.get("FMA_BIND_HOST", "0.0.0.0")
    return args


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker_entry(rank: int, world: int, master_port: int,
                  model_name: str, max_model_len: Optional[int],
                  seed: int, expert_parallel: bool = False) -> None:
    # ranks 1..N-1: build the engine shard and serve TP commands forever
    # (workers stay in the instance's process group so the launcher's
    # killpg on force-stop reaps them too)
    from fma_amd.parallel import tp
    from fma_amd.runtime.engine import ActuationEngine

    ctx = tp.init_tp(rank, world, master_port, list(range(world)))
    checkpoint = model_name if os.path.isdir(model_name) else None
    if checkpoint:
        from fma_amd.models import loader
        cfg = loader.config_from_dir(checkpoint) or LlamaConfig.tiny()
    else:
        cfg = LlamaConfig.by_name(model_name)
    if max_model_len:
        cfg.max_seq_len = max_model_len
    if expert_parallel:
        cfg.expert_parallel = True
    eng = ActuationEngine(cfg, device_index=ctx.device_index,
                          tp_rank=rank, tp_size=world,
                          tp_group=ctx.device_group, seed=seed,
                          init_weights=checkpoint is None)
    if checkpoint:
        eng.load_checkpoint(checkpoint)
    tp.run_worker_loop(ctx, eng)


def load_eos_id(checkpoint_dir: str):
    """eos_token_id from the HF config.json (first of a list), so text
    completions stop at end-of-sequence like the upstream model."""
    import json as _json
    path = os.path.join(checkpoint_dir, "config.json")
    try:
        with open(path) as f:
            eos = _json.load(f).get("eos_token_id")
        if isinstance(eos, list):
            eos = eos[0] if eos else None
        return int(eos) if eos is not None else None
    except (OSError, ValueError, TypeError):
        return None


def load_tokenizer(checkpoint_dir: str):
    """HF checkpoint dirs carry tokenizer.json; attach the real
    tokenizer so /v1/completions speaks text, not byte ids. Returns
    None (byte-level fallback) when absent or unloadable."""
    path = os.path.join(checkpoint_dir, "tokenizer.json")
    if not os.path.exists(path):
        return None
    try:
        from tokenizers import Tokenizer
        return Tokenizer.from_file(path)
    except Exception as e:  # noqa: BLE001 - degraded, not fatal
        print(f"[serve] tokenizer load failed ({e}); byte-level fallback",
              flush=True)
        return None


class ServingRuntime:
    """Everything behind the HTTP app: engine (+TP fan-out) and state."""

    def __init__(self, args: argparse.Namespace):
        from fma_amd.parallel import tp
        from fma_amd.runtime.engine import ActuationEngine

        self.args = args
        self.lock = threading.Lock()
        self.started_at = time.time()
        world = args.tensor_parallel_size
        self.workers: List[multiprocessing.Process] = []
        master_port = _free_port() if world > 1 else 0
        mp = multiprocessing.get_context("fork")
        for r in range(1, world):
            p = mp.Process(target=_worker_entry,
                           args=(r, world, master_port, args.model,
                                 args.max_model_len, args.seed,
                                 getattr(args, "expert_parallel", False)),
                           daemon=True)
            p.start()
            self.workers.append(p)
        ctx = tp.init_tp(0, world, master_port, list(range(world)))
        checkpoint = None
        if os.path.isdir(args.model):
            from fma_amd.models import loader
            checkpoint = args.model
            cfg = loader.config_from_dir(checkpoint) or LlamaConfig.tiny()
            # TP > 1: each rank slices its Megatron shard out of the full
            # checkpoint (loader.shard_slice); workers load their own in
            # _worker_entry before entering the command loop
        else:
            cfg = LlamaConfig.by_name(args.model)
        if args.max_model_len:
            cfg.max_seq_len = args.max_model_len
        if getattr(args, "expert_parallel", False):
            cfg.expert_parallel = True
        engine = ActuationEngine(cfg, device_index=ctx.device_index,
                                 tp_rank=0, tp_size=world,
                                 tp_group=ctx.device_group, seed=args.seed,
                                 init_weights=checkpoint is None)
        if checkpoint is not None:
            t_load = engine.load_checkpoint(checkpoint)
            print(f"[serve] checkpoint {checkpoint} loaded in {t_load:.2f}s",
                  flush=True)
            engine.tokenizer = load_tokenizer(checkpoint)
            engine.eos_id = load_eos_id(checkpoint)
        self.rt = tp.TPRuntime(ctx, engine) if world > 1 else engine
        self.model_name = args.served_model_name or args.model
        print(f"[serve] engine up in {engine.create_seconds:.2f}s "
              f"({engine.total_bytes/2**30:.2f} GiB, "
              f"pid={os.getpid()}, world={world}, "
              f"timing={getattr(engine, 'timing', {})})", flush=True)
        if args.start_asleep:
            self.rt.sleep(1)
        self.ready = True

    def close(self) -> None:
        if hasattr(self.rt, "stop"):
            self.rt.stop()
        for p in self.workers:
            p.join(timeout=5)


def create_app(runtime: ServingRuntime) -> FastAPI:
    app = FastAPI(title="fma-amd inference server", version="0.1")
    rt = runtime.rt

    @app.get("/health")
    def health():
        return {"status": "OK"}

    @app.get("/is_sleeping")
    def is_sleeping():
        # JSON shape per reference pkg/api/interface.go:131-135
        return {"is_sleeping": rt.is_sleeping()}

    @app.post("/sleep")
    def sleep(level: int = Query(default=1)):
        with runtime.lock:
            t = rt.sleep(level)
        return {"status": "ok", "level": level, "seconds": t}

    @app.post("/wake_up")
    def wake_up():
        with runtime.lock:
            t = rt.wake_up()
        return {"status": "ok", "seconds": t}

    @app.get("/v1/models")
    def models():
        return {"object": "list", "data": [{
            "id": runtime.model_name, "object": "model",
            "owned_by": "fma-amd",
        }]}

    @app.post("/v1/completions")
    def completions(body: dict):
        prompt = body.get("prompt", "")
        max_tokens = int(body.get("max_tokens", 16))
        with runtime.lock:
            if rt.is_sleeping():
                return JSONResponse(
                    {"error": "model is sleeping"}, status_code=409)
            t0 = time.time()
            temperature = float(body.get("temperature", 0.0))
            top_p = float(body.get("top_p", 1.0))
            if isinstance(prompt, list):
                # OpenAI-compat: prompt may be token ids; respond with
                # the generated ids so clients control detokenization
                import torch as _torch
                eng = rt.engine if hasattr(rt, "engine") else rt
                ids = [int(t) % eng.cfg.vocab_size for t in prompt] or [1]
                toks = _torch.tensor([ids], dtype=_torch.long,
                                     device=eng.device)
                kw = ({"temperature": temperature, "top_p": top_p}
                      if temperature > 0 and not hasattr(rt, "ctx")
                      else {})  # TP ranks must agree: greedy only
                out = rt.generate(toks, max_tokens, **kw)[0, len(ids):]
                return {
                    "id": f"cmpl-{int(t0*1e6)}",
                    "object": "text_completion",
                    "model": runtime.model_name,
                    "choices": [{"index": 0,
                                 "token_ids": [int(t) for t in out],
                                 "text": "",
                                 "finish_reason": "length"}],
                    "usage": {"prompt_tokens": len(ids),
                              "completion_tokens": max_tokens},
                }
            text = rt.generate_text(prompt, max_tokens,
                                    temperature=temperature, top_p=top_p) \
                if not hasattr(rt, "ctx") else \
                rt.generate_text(prompt, max_tokens)
        return {
            "id": f"cmpl-{int(t0*1e6)}",
            "object": "text_completion",
            "model": runtime.model_name,
            "choices": [{"index": 0, "text": text,
                         "finish_reason": "length"}],
            "usage": {"completion_tokens": max_tokens},
        }

    @app.post("/v1/chat/completions")
    def chat_completions(body: dict):
        """Minimal chat surface (vLLM serves it; routers probe it):
        messages are rendered with the reversible generic template
        "<|role|>content" per turn + a final assistant header. Models
        with their own chat template should be driven through
        /v1/completions with pre-templated text."""
        msgs = body.get("messages") or []
        max_tokens = int(body.get("max_tokens", 16))
        temperature = float(body.get("temperature", 0.0))
        top_p = float(body.get("top_p", 1.0))
        prompt = "".join(
            f"<|{m.get('role', 'user')}|>{m.get('content', '')}"
            for m in msgs) + "<|assistant|>"
        with runtime.lock:
            if rt.is_sleeping():
                return JSONResponse(
                    {"error": "model is sleeping"}, status_code=409)
            t0 = time.time()
            kw = ({"temperature": temperature, "top_p": top_p}
                  if not hasattr(rt, "ctx") else {})
            text = rt.generate_text(prompt, max_tokens, **kw)
        return {
            "id": f"chatcmpl-{int(t0*1e6)}",
            "object": "chat.completion",
            "model": runtime.model_name,
            "choices": [{"index": 0,
                         "message": {"role": "assistant",
                                     "content": text},
                         "finish_reason": "length"}],
            "usage": {"completion_tokens": max_tokens},
        }

    @app.get("/stats")
    def stats():
        return rt.stats()

    @app.get("/metrics")
    def metrics():
        """Prometheus exposition (vLLM serves /metrics too): engine
        counters + last actuation timings as gauges, served without a
        global registry so multiple instances coexist in one process."""
        st = rt.stats()
        lines = []

        def g(name, val, help_):
            if val is None:
                return
            lines.append(f"# HELP {name} {help_}")
            lines.append(f"# TYPE {name} gauge")
            lines.append(f"{name} {float(val)}")

        g("fma_engine_sleep_count", st.get("sleep_count"),
          "sleep() calls since instance start")
        g("fma_engine_wake_count", st.get("wake_count"),
          "wake_up() calls since instance start")
        g("fma_engine_last_sleep_seconds", st.get("last_sleep_seconds"),
          "duration of the most recent sleep")
        g("fma_engine_last_wake_seconds", st.get("last_wake_seconds"),
          "duration of the most recent wake_up")
        g("fma_engine_param_bytes", st.get("param_bytes"),
          "model parameter bytes resident when awake")
        g("fma_engine_is_sleeping", 1.0 if rt.is_sleeping() else 0.0,
          "1 while weights are offloaded to host DRAM")
        from fastapi.responses import PlainTextResponse
        return PlainTextResponse("\n".join(lines) + "\n",
                                 media_type="text/plain; version=0.0.4")

    return app


def main(argv: Optional[List[str]] = None) -> None:
    import sys

    import uvicorn

    t0 = time.time()
    args = parse_options(" ".join(argv if argv is not None else sys.argv[1:]))
    print(f"[serve] parsing done at +{time.time()-t0:.2f}s", flush=True)
    runtime = ServingRuntime(args)
    print(f"[serve] runtime ready at +{time.time()-t0:.2f}s", flush=True)
    app = create_app(runtime)
    try:
        uvicorn.run(app, host=args.host, port=args.port, log_level="warning")
    finally:
        runtime.close()


if __name__ == "__main__":
    main()
