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


CKPT_PROBE = r"""
import os
os.environ.setdefault("FMA_FAKE_GPU", "1")
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
import sys
sys.path.insert(0, "__ROOT__")
import torch
from fma_amd.runtime.server import ServingRuntime, parse_options

# TP=2 runtime loads a full (unsharded) checkpoint, each rank slicing
# its Megatron shard; greedy decode must be numerically identical to
# the single-process full model (reference output precomputed by the
# test — engines must NOT be created pre-fork in this process).
ref = torch.load(os.path.join("__TMP__", "ref.pt"), weights_only=True)
toks = torch.load(os.path.join("__TMP__", "toks.pt"), weights_only=True)
rt = ServingRuntime(parse_options(
    "--model " + os.path.join("__TMP__", "tp-ckpt")
    + " --tensor-parallel-size 2 --seed 4"))
out = rt.rt.generate(toks, max_new_tokens=3)
assert torch.equal(out, ref), (out, ref)
rt.rt.stop()
print("TP_CKPT_OK")
"""


def test_tp_checkpoint_sharded_load_matches_full_model(tmp_path):
    """TP=2 serving from a full checkpoint (per-rank Megatron slicing)
    decodes token-identically to the single-process full model."""
    import torch
    from fma_amd.models import loader
    from fma_amd.models.llama import LlamaConfig
    from fma_amd.runtime.engine import ActuationEngine

    os.environ.setdefault("FMA_FAKE_GPU", "1")
    cfg = LlamaConfig(name="tpck", vocab_size=64, hidden_size=64,
                      intermediate_size=96, num_layers=2, num_heads=4,
                      num_kv_heads=2, max_seq_len=32)
    src = ActuationEngine(cfg, seed=17)
    loader.save_params(src.params, str(tmp_path / "tp-ckpt"), cfg)
    toks = torch.randint(0, cfg.vocab_size, (1, 5),
                         generator=torch.Generator().manual_seed(3))
    torch.save(toks, str(tmp_path / "toks.pt"))
    torch.save(src.model.generate(toks, max_new_tokens=3),
               str(tmp_path / "ref.pt"))

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    probe = CKPT_PROBE.replace("__ROOT__", root).replace(
        "__TMP__", str(tmp_path))
    res = subprocess.run(
        [sys.executable, "-c", probe],
        capture_output=True, text=True, timeout=150,
        env=dict(os.environ, PYTHONPATH=root))
    assert res.returncode == 0, f"stdout={res.stdout}\nstderr={res.stderr}"
    assert "TP_CKPT_OK" in res.stdout


KVREP_PROBE = r"""
# tp_size 4 > num_kv_heads 2: each pair of ranks REPLICATES one logical
# kv head (Megatron grouping); decode must still match the full model.
import os
os.environ.setdefault("FMA_FAKE_GPU", "1")
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
import sys
sys.path.insert(0, "__ROOT__")
import torch
from fma_amd.runtime.server import ServingRuntime, parse_options

ref = torch.load(os.path.join("__TMP__", "ref.pt"), weights_only=True)
toks = torch.load(os.path.join("__TMP__", "toks.pt"), weights_only=True)
rt = ServingRuntime(parse_options(
    "--model " + os.path.join("__TMP__", "kvrep-ckpt")
    + " --tensor-parallel-size 4 --seed 4"))
out = rt.rt.generate(toks, max_new_tokens=3)
assert torch.equal(out, ref), (out, ref)
# sleep/wake through the replicated shards stays bit-stable
before = rt.rt.generate(toks, max_new_tokens=2).clone()
rt.rt.sleep(1)
rt.rt.wake_up()
assert torch.equal(rt.rt.generate(toks, max_new_tokens=2), before)
rt.rt.stop()
print("TP_KVREP_OK")
"""


def test_tp_exceeding_kv_heads_replicates(tmp_path):
    """tp_size > num_kv_heads (VERDICT round-1 gap): ranks sharing a
    logical kv head load identical replicated wk/wv slices, and TP=4
    decode over kv_heads=2 is token-identical to the full model."""
    import torch
    from fma_amd.models import loader
    from fma_amd.models.llama import LlamaConfig
    from fma_amd.runtime.engine import ActuationEngine

    os.environ.setdefault("FMA_FAKE_GPU", "1")
    cfg = LlamaConfig(name="kvrep", vocab_size=64, hidden_size=64,
                      intermediate_size=96, num_layers=2, num_heads=4,
                      num_kv_heads=2, max_seq_len=32)
    src = ActuationEngine(cfg, seed=23)
    loader.save_params(src.params, str(tmp_path / "kvrep-ckpt"), cfg)
    toks = torch.randint(0, cfg.vocab_size, (1, 5),
                         generator=torch.Generator().manual_seed(5))
    torch.save(toks, str(tmp_path / "toks.pt"))
    torch.save(src.model.generate(toks, max_new_tokens=3),
               str(tmp_path / "ref.pt"))

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    probe = KVREP_PROBE.replace("__ROOT__", root).replace(
        "__TMP__", str(tmp_path))
    res = subprocess.run(
        [sys.executable, "-c", probe],
        capture_output=True, text=True, timeout=150,
        env=dict(os.environ, PYTHONPATH=root))
    assert res.returncode == 0, f"stdout={res.stdout}\nstderr={res.stderr}"
    assert "TP_KVREP_OK" in res.stdout


def test_shard_slice_kv_replication_unit():
    import torch
    from fma_amd.models.loader import shard_slice

    full = torch.arange(4 * 8, dtype=torch.float32).view(4, 8)  # 2 heads*2
    # tp=4, kv rows 4 (2 heads of dim0=2): local_rows=2 per rank,
    # ranks 0,1 -> head 0 rows, ranks 2,3 -> head 1 rows
    for r, expect_lo in [(0, 0), (1, 0), (2, 2), (3, 2)]:
        got = shard_slice("layers.0.wk.weight", full, r, 4, local_rows=2)
        assert torch.equal(got, full[expect_lo:expect_lo + 2]), r
    # without replication info, non-divisible raises
    import pytest as _pytest
    with _pytest.raises(ValueError):
        shard_slice("layers.0.wk.weight", full, 0, 3)


MOE_PROBE = r"""
# Mixtral-style MoE under TP=2: experts Megatron-sharded, router
# replicated; decode must match the full model token for token.
import os
os.environ.setdefault("FMA_FAKE_GPU", "1")
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
import sys
sys.path.insert(0, "__ROOT__")
import torch
from fma_amd.runtime.server import ServingRuntime, parse_options

ref = torch.load(os.path.join("__TMP__", "ref.pt"), weights_only=True)
toks = torch.load(os.path.join("__TMP__", "toks.pt"), weights_only=True)
rt = ServingRuntime(parse_options(
    "--model " + os.path.join("__TMP__", "moe-ckpt")
    + " --tensor-parallel-size 2 --seed 4"))
out = rt.rt.generate(toks, max_new_tokens=3)
assert torch.equal(out, ref), (out, ref)
before = rt.rt.generate(toks, max_new_tokens=2).clone()
rt.rt.sleep(1)
rt.rt.wake_up()
assert torch.equal(rt.rt.generate(toks, max_new_tokens=2), before)
rt.rt.stop()
print("TP_MOE_OK")
"""


def test_tp_moe_checkpoint_matches_full_model(tmp_path):
    """MoE TP=2 from a full checkpoint (per-expert Megatron slicing)
    decodes token-identically to the single-process full model."""
    import torch
    from fma_amd.models import loader
    from fma_amd.models.llama import LlamaConfig
    from fma_amd.runtime.engine import ActuationEngine

    os.environ.setdefault("FMA_FAKE_GPU", "1")
    cfg = LlamaConfig(name="moeck", vocab_size=64, hidden_size=64,
                      intermediate_size=96, num_layers=2, num_heads=4,
                      num_kv_heads=2, max_seq_len=32, num_experts=4,
                      num_experts_per_tok=2)
    src = ActuationEngine(cfg, seed=31)
    loader.save_params(src.params, str(tmp_path / "moe-ckpt"), cfg)
    toks = torch.randint(0, cfg.vocab_size, (1, 5),
                         generator=torch.Generator().manual_seed(9))
    torch.save(toks, str(tmp_path / "toks.pt"))
    torch.save(src.model.generate(toks, max_new_tokens=3),
               str(tmp_path / "ref.pt"))

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    probe = MOE_PROBE.replace("__ROOT__", root).replace(
        "__TMP__", str(tmp_path))
    res = subprocess.run(
        [sys.executable, "-c", probe],
        capture_output=True, text=True, timeout=150,
        env=dict(os.environ, PYTHONPATH=root))
    assert res.returncode == 0, f"stdout={res.stdout}\nstderr={res.stderr}"
    assert "TP_MOE_OK" in res.stdout


EP_PROBE = r"""
# Expert parallelism: whole experts partitioned across 2 ranks
# (attention Megatron-TP); decode must match the full model.
import os
os.environ.setdefault("FMA_FAKE_GPU", "1")
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
import sys
sys.path.insert(0, "__ROOT__")
import torch
from fma_amd.runtime.server import ServingRuntime, parse_options

ref = torch.load(os.path.join("__TMP__", "ref.pt"), weights_only=True)
toks = torch.load(os.path.join("__TMP__", "toks.pt"), weights_only=True)
rt = ServingRuntime(parse_options(
    "--model " + os.path.join("__TMP__", "moe-ckpt")
    + " --tensor-parallel-size 2 --expert-parallel --seed 4"))
eng = rt.rt.engine
# rank 0 holds the first half of the experts, whole
assert "layers.0.experts.0.w_gate.weight" in eng.params
assert "layers.0.experts.3.w_gate.weight" not in eng.params
assert eng.params["layers.0.experts.0.w_gate.weight"].shape[0] == 96
out = rt.rt.generate(toks, max_new_tokens=3)
assert torch.equal(out, ref), (out, ref)
before = rt.rt.generate(toks, max_new_tokens=2).clone()
rt.rt.sleep(1)
rt.rt.wake_up()
assert torch.equal(rt.rt.generate(toks, max_new_tokens=2), before)
rt.rt.stop()
print("TP_EP_OK")
"""


def test_expert_parallel_matches_full_model(tmp_path):
    """EP=2 over a full MoE checkpoint: each rank loads only its whole
    experts, the per-layer all-reduce sums routed outputs, and greedy
    decode is token-identical to the single-process full model."""
    import torch
    from fma_amd.models import loader
    from fma_amd.models.llama import LlamaConfig
    from fma_amd.runtime.engine import ActuationEngine

    os.environ.setdefault("FMA_FAKE_GPU", "1")
    cfg = LlamaConfig(name="epck", vocab_size=64, hidden_size=64,
                      intermediate_size=96, num_layers=2, num_heads=4,
                      num_kv_heads=2, max_seq_len=32, num_experts=4,
                      num_experts_per_tok=2)
    src = ActuationEngine(cfg, seed=41)
    loader.save_params(src.params, str(tmp_path / "moe-ckpt"), cfg)
    toks = torch.randint(0, cfg.vocab_size, (1, 5),
                         generator=torch.Generator().manual_seed(11))
    torch.save(toks, str(tmp_path / "toks.pt"))
    torch.save(src.model.generate(toks, max_new_tokens=3),
               str(tmp_path / "ref.pt"))

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    probe = EP_PROBE.replace("__ROOT__", root).replace(
        "__TMP__", str(tmp_path))
    res = subprocess.run(
        [sys.executable, "-c", probe],
        capture_output=True, text=True, timeout=150,
        env=dict(os.environ, PYTHONPATH=root))
    assert res.returncode == 0, f"stdout={res.stdout}\nstderr={res.stderr}"
    assert "TP_EP_OK" in res.stdout
