"""TP serving runtime on CPU (gloo, world_size 2, fork workers).

Runs in a fresh subprocess: forking TP workers from a long-lived pytest
process (which accumulates helper threads from other test modules) can
inherit held locks; a clean interpreter is the supported spawn context —
the launcher always forks instances from a dedicated parent process.
"""

import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.timeout(180)

PROBE = r"""
import os
os.environ.setdefault("FMA_FAKE_GPU", "1")
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
import sys
sys.path.insert(0, "__ROOT__")
import torch
from fma_amd.runtime.server import ServingRuntime, parse_options

rt = ServingRuntime(parse_options(
    "--model tiny --tensor-parallel-size 2 --seed 4"))
r = rt.rt
assert not r.is_sleeping()
toks = torch.randint(0, r.engine.cfg.vocab_size, (1, 6))
# NOTE: every model invocation must go through the TPRuntime command API
# (a bare rank-0 forward would issue collectives with no partner)
before = r.generate(toks, max_new_tokens=3).clone()
param_snap = {n: p.clone() for n, p in r.engine.params.items()}
r.sleep(1)
assert r.is_sleeping()
r.wake_up()  # includes the all-rank barrier
assert not r.is_sleeping()
for n, p in r.engine.params.items():
    assert torch.equal(p, param_snap[n]), f"rank0 param {n} corrupted"
after = r.generate(toks, max_new_tokens=3)
assert torch.equal(before, after), "TP sleep/wake changed decode output"
assert after.shape == (1, 9), after.shape
assert r.stats()["tp_size"] == 2
txt = r.generate_text("hello", 2)
assert isinstance(txt, str)
r.stop()
print("TP_PROBE_OK")
"""


def test_tp_runtime_sleep_wake_generate(tmp_path):
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    res = subprocess.run(
        [sys.executable, "-c", PROBE.replace("__ROOT__", root)],
        capture_output=True, text=True, timeout=150,
        env=dict(os.environ, PYTHONPATH=root))
    assert res.returncode == 0, f"stdout={res.stdout}\nstderr={res.stderr}"
    assert "TP_PROBE_OK" in res.stdout
