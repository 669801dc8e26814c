"""bench.py IS the driver contract: one JSON line on stdout with the
exact headline metric, runnable standalone and under torchrun (the
driver launches N>1 exactly that way). Runs here on CPU via the fake
arena; the same code path moves real HBM bytes on a GPU box."""

import json
import os
import socket
import subprocess
import sys

import pytest

pytestmark = pytest.mark.timeout(300)

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def run_bench(cmd, timeout=240):
    env = dict(os.environ, FMA_FAKE_GPU="1", PYTHONPATH=ROOT,
               MASTER_ADDR="127.0.0.1")
    res = subprocess.run(cmd, capture_output=True, text=True,
                         timeout=timeout, env=env, cwd=ROOT)
    assert res.returncode == 0, f"stdout={res.stdout}\nstderr={res.stderr}"
    # gloo prints its connection banner to stdout on CPU; RCCL on the
    # GPU box does not (without NCCL_DEBUG) — the contract is exactly
    # one JSON line, so: exactly one line must parse as JSON
    parsed = []
    for ln in res.stdout.splitlines():
        ln = ln.strip()
        if ln.startswith("{"):
            parsed.append(json.loads(ln))
    assert len(parsed) == 1, f"expected ONE json line, got: {res.stdout!r}"
    return parsed[0]


def test_bench_single_process_contract():
    out = run_bench([sys.executable, "bench.py", "--steps", "2",
                     "--warmup", "1", "--gib", "0.05"])
    assert REQUIRED <= set(out)
    assert out["metric"].startswith("wake_up latency (s)")
    assert out["n_gpus"] == 1 and out["steps"] == 2 and out["warmup"] == 1
    assert out["higher_is_better"] is False
    assert out["scaling"] == "strong"
    assert out["unit"] == "s"
    assert out["value"] > 0 and out["ms_per_step"] > 0
    assert out["dtype"] == "bf16"
    assert "synthetic" in out["data"]
    assert out["vs_baseline"] is None  # only quoted at the named 64 GiB
    assert out["config"]["parallelism"] == "tp1"


def test_bench_torchrun_world2_gloo():
    """The exact launch shape the driver uses for N>1 (gloo here; RCCL
    on the GPU box): one rank per GPU, MAX over ranks, rank 0 prints."""
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    out = run_bench([
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
        "--master-port", str(port), "bench.py", "--gpus", "2",
        "--steps", "1", "--warmup", "0", "--gib", "0.05"])
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "tp2"
    # strong scaling: per-rank shard is about half the total
    assert out["config"]["param_gib_per_rank"] < 0.05
