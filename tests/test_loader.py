"""Safetensors checkpoint save/load into arena-backed engines."""

import os

import pytest
import torch

os.environ.setdefault("FMA_FAKE_GPU", "1")

from fma_amd.models import loader  # noqa: E402
from fma_amd.models.llama import LlamaConfig  # noqa: E402
from fma_amd.runtime.engine import ActuationEngine  # noqa: E402


def test_save_load_roundtrip(tmp_path):
    src = ActuationEngine(LlamaConfig.tiny(), seed=21)
    ckpt = str(tmp_path / "tiny-ckpt")
    loader.save_params(src.params, ckpt, src.cfg)

    assert loader.premap_safetensors(ckpt) > 0
    cfg = loader.config_from_dir(ckpt)
    assert cfg.hidden_size == src.cfg.hidden_size

    dst = ActuationEngine(cfg, seed=99, init_weights=False)
    n = loader.load_into_params(ckpt, dst.params)
    assert n == len(src.params)
    toks = torch.randint(0, cfg.vocab_size, (1, 5))
    assert torch.equal(src.model.forward(toks), dst.model.forward(toks))

    # sleep/wake of a checkpoint-loaded engine is still bit-exact
    before = dst.model.forward(toks).clone()
    dst.sleep()
    dst.wake_up()
    assert torch.equal(before, dst.model.forward(toks))


def test_load_strict_mismatch(tmp_path):
    src = ActuationEngine(LlamaConfig.tiny(), seed=1)
    ckpt = str(tmp_path / "ck")
    loader.save_params(src.params, ckpt, src.cfg)
    other = ActuationEngine(LlamaConfig.tiny(), seed=2)
    del other.params["final_norm.weight"]
    with pytest.raises(KeyError):
        loader.load_into_params(ckpt, other.params)


def test_server_loads_checkpoint_dir(tmp_path):
    from fma_amd.runtime.server import ServingRuntime, parse_options
    src = ActuationEngine(LlamaConfig.tiny(), seed=5)
    ckpt = str(tmp_path / "m")
    loader.save_params(src.params, ckpt, src.cfg)
    rt = ServingRuntime(parse_options(f"--model {ckpt}"))
    toks = torch.randint(0, src.cfg.vocab_size, (1, 4))
    assert torch.equal(src.model.forward(toks),
                       rt.rt.model.forward(toks))


def test_fast_load_checkpoint_into_arena(tmp_path):
    src = ActuationEngine(LlamaConfig.tiny(), seed=31)
    ckpt = str(tmp_path / "fast-ckpt")
    loader.save_params(src.params, ckpt, src.cfg)
    dst = ActuationEngine(LlamaConfig.tiny(), seed=77, init_weights=False)
    t = dst.load_checkpoint(ckpt)
    assert t >= 0
    toks = torch.randint(0, src.cfg.vocab_size, (1, 6))
    assert torch.equal(src.model.forward(toks), dst.model.forward(toks))
    # still sleep/wake clean afterwards
    before = dst.model.forward(toks).clone()
    dst.sleep()
    dst.wake_up()
    assert torch.equal(before, dst.model.forward(toks))


def test_fast_load_missing_param_raises(tmp_path):
    src = ActuationEngine(LlamaConfig.tiny(), seed=1)
    ckpt = str(tmp_path / "partial")
    params = dict(src.params)
    params.pop("final_norm.weight")
    loader.save_params(params, ckpt, src.cfg)
    dst = ActuationEngine(LlamaConfig.tiny(), seed=2, init_weights=False)
    import pytest as _pytest
    with _pytest.raises(KeyError):
        dst.load_checkpoint(ckpt)
