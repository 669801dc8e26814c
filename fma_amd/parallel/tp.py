"""Tensor-parallel process management for a serving instance.

The reference treats TP as an opaque ``--tensor-parallel-size`` option
forwarded to vLLM (reference docs/launcher.md:584-595); here it is native:
one process per GPU, ``torch.distributed`` with the nccl backend (RCCL over
xGMI on ROCm) for model collectives and the wake barrier, plus a gloo
sub-group for control-plane object broadcast (commands flow rank0 ->
workers without touching the GPU).

Process model: the instance process (spawned by the launcher) becomes TP
rank 0, serves HTTP, and forks ranks 1..N-1 running :func:`worker_main`.
Every command (sleep / wake_up / generate) executes SPMD on all ranks; the
RCCL barrier inside ``ActuationEngine.wake_up`` guarantees all ranks have
re-materialized their shard before rank 0 reports ``is_sleeping=false``.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

import torch
import torch.distributed as dist

CMD_SLEEP = "sleep"
CMD_WAKE = "wake_up"
CMD_GENERATE = "generate"
CMD_STOP = "stop"


@dataclass
class TPContext:
    rank: int
    world: int
    device_index: int
    ctrl_group: Optional[object]  # gloo, for object broadcast
    device_group: Optional[object]  # nccl(RCCL) on GPU, gloo on CPU


def init_tp(rank: int, world: int, master_port: int,
            device_indices: List[int]) -> TPContext:
    """Initialize the process group for one TP rank.

    Always rendezvous on 127.0.0.1 (single node; xGMI links are intra-node).
    """
    if world <= 1:
        return TPContext(0, 1, device_indices[0] if device_indices else 0,
                         None, None)
    on_gpu = torch.cuda.is_available()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(master_port)
    backend = "nccl" if on_gpu else "gloo"
    device_index = device_indices[rank] if rank < len(device_indices) else rank
    if on_gpu:
        torch.cuda.set_device(device_index)
    dist.init_process_group(backend=backend, rank=rank, world_size=world)
    device_group = dist.group.WORLD
    # dedicated gloo group for command objects (never blocks on the GPU)
    ctrl_group = dist.new_group(backend="gloo") if backend != "gloo" \
        else device_group
    return TPContext(rank, world, device_index, ctrl_group, device_group)


def broadcast_cmd(ctx: TPContext, cmd: Optional[Dict[str, Any]] = None
                  ) -> Dict[str, Any]:
    """rank0 sends `cmd`; workers pass None and receive it."""
    if ctx.world <= 1:
        return cmd or {}
    buf: List[Any] = [cmd]
    dist.broadcast_object_list(buf, src=0, group=ctx.ctrl_group)
    return buf[0]


def run_worker_loop(ctx: TPContext, engine) -> None:
    """Ranks 1..N-1: execute commands until STOP."""
    while True:
        cmd = broadcast_cmd(ctx)
        op = cmd.get("op")
        if op == CMD_STOP:
            break
        if op == CMD_SLEEP:
            engine.sleep(cmd.get("level", 1))
        elif op == CMD_WAKE:
            engine.wake_up()
        elif op == CMD_GENERATE:
            tokens = torch.tensor(cmd["tokens"], dtype=torch.long,
                                  device=engine.device)
            engine.generate(tokens, cmd.get("max_new_tokens", 16))
        else:  # pragma: no cover - protocol error
            raise RuntimeError(f"unknown TP command {op!r}")
    if dist.is_initialized():
        dist.destroy_process_group()


class TPRuntime:
    """Rank-0 view of a TP instance: mirrors the engine API but drives all
    ranks through the control group before executing locally."""

    def __init__(self, ctx: TPContext, engine):
        self.ctx = ctx
        self.engine = engine

    def is_sleeping(self) -> bool:
        return self.engine.is_sleeping()

    def sleep(self, level: int = 1) -> float:
        broadcast_cmd(self.ctx, {"op": CMD_SLEEP, "level": level})
        return self.engine.sleep(level)

    def wake_up(self) -> float:
        broadcast_cmd(self.ctx, {"op": CMD_WAKE})
        return self.engine.wake_up()

    def generate(self, tokens: torch.Tensor, max_new_tokens: int = 16):
        broadcast_cmd(self.ctx, {
            "op": CMD_GENERATE,
            "tokens": tokens.tolist(),
            "max_new_tokens": max_new_tokens,
        })
        return self.engine.generate(tokens, max_new_tokens)

    def generate_text(self, prompt: str, max_new_tokens: int = 16,
                      temperature: float = 0.0, top_p: float = 1.0) -> str:
        # TP ranks must agree on every sampled token; greedy only
        tok = getattr(self.engine, "tokenizer", None)
        if tok is not None:
            ids = tok.encode(prompt).ids or [1]
            toks = torch.tensor([ids], dtype=torch.long,
                                device=self.engine.device)
            out = self.generate(toks, max_new_tokens)[0, len(ids):]
            return tok.decode([int(t) for t in out.tolist()])
        ids = [b % self.engine.cfg.vocab_size for b in prompt.encode("utf-8")] \
            or [1]
        toks = torch.tensor([ids], dtype=torch.long, device=self.engine.device)
        out = self.generate(toks, max_new_tokens)[0, len(ids):]
        return bytes(int(t) % 256 for t in out.tolist()).decode(
            "utf-8", errors="replace")

    def stats(self) -> Dict[str, Any]:
        s = self.engine.stats()
        s["tp_size"] = self.ctx.world
        return s

    def stop(self) -> None:
        broadcast_cmd(self.ctx, {"op": CMD_STOP})
        if dist.is_initialized():
            dist.destroy_process_group()
