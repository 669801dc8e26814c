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


# -- TP-sharded checkpoint loading -------------------------------------------

def _tp_cfg():
    return LlamaConfig(name="tpc", vocab_size=64, hidden_size=64,
                       intermediate_size=96, num_layers=2, num_heads=4,
                       num_kv_heads=2, max_seq_len=32)


def test_shard_slice_reassembles_full_tensor():
    torch.manual_seed(5)
    full = {
        "layers.0.wq.weight": torch.randn(16, 8),      # col-parallel (dim0)
        "layers.0.wo.weight": torch.randn(8, 16),      # row-parallel (dim1)
        "layers.0.attn_norm.weight": torch.randn(8),   # replicated
        "embed.weight": torch.randn(10, 8),            # replicated
    }
    for name, t in full.items():
        shards = [loader.shard_slice(name, t, r, 2) for r in range(2)]
        if "wq" in name:
            assert torch.equal(torch.cat(shards, dim=0), t)
        elif "wo" in name:
            assert torch.equal(torch.cat(shards, dim=1), t)
        else:
            assert all(torch.equal(s, t) for s in shards)


def test_shard_slice_rejects_indivisible():
    import pytest as _pytest
    with _pytest.raises(ValueError):
        loader.shard_slice("layers.0.wk.weight", torch.randn(3, 8), 0, 2)


def test_load_into_params_tp_sharded(tmp_path):
    """A full checkpoint loads into each rank's Megatron shard; the
    shards stitched back together equal the full parameters."""
    cfg = _tp_cfg()
    src = ActuationEngine(cfg, seed=9)
    ckpt = str(tmp_path / "full-ckpt")
    loader.save_params(src.params, ckpt, cfg)

    col = {"wq", "wk", "wv", "w_gate", "w_up"}
    row = {"wo", "w_down"}
    for tp_size in (2,):
        shards = []
        for r in range(tp_size):
            params = {n: torch.zeros(s, dtype=d)
                      for n, s, d in cfg.param_specs(r, tp_size)}
            n = loader.load_into_params(ckpt, params,
                                        tp_rank=r, tp_size=tp_size)
            assert n == len(params)
            shards.append(params)
        for name, fullp in src.params.items():
            leaf = name.split(".")[-2]
            if leaf in col:
                glued = torch.cat([s[name] for s in shards], dim=0)
            elif leaf in row:
                glued = torch.cat([s[name] for s in shards], dim=1)
            else:
                glued = shards[0][name]
            assert torch.equal(glued, fullp), name


def test_engine_tp_rank_load_checkpoint(tmp_path):
    """An engine constructed as TP rank 1 of 2 loads its shard straight
    from a full checkpoint through the arena staging path."""
    cfg = _tp_cfg()
    src = ActuationEngine(cfg, seed=13)
    ckpt = str(tmp_path / "full2")
    loader.save_params(src.params, ckpt, cfg)
    eng = ActuationEngine(cfg, seed=1, tp_rank=1, tp_size=2,
                          init_weights=False)
    eng.load_checkpoint(ckpt)
    q_local = cfg.num_heads // 2 * cfg.head_dim
    assert torch.equal(eng.params["layers.0.wq.weight"],
                       src.params["layers.0.wq.weight"][q_local:])
    i_local = cfg.intermediate_size // 2
    assert torch.equal(eng.params["layers.1.w_down.weight"],
                       src.params["layers.1.w_down.weight"][:, i_local:])
    assert torch.equal(eng.params["final_norm.weight"],
                       src.params["final_norm.weight"])


def test_config_roundtrip_preserves_family_fields(tmp_path):
    """config.json must carry the family-defining fields (qkv_bias, MoE
    shape, norm_eps) — a dropped field silently loads a different
    architecture."""
    import torch

    from fma_amd.models import loader
    from fma_amd.models.llama import LlamaConfig

    for preset in ("tiny", "tiny-qwen", "tiny-moe"):
        cfg = LlamaConfig.by_name(preset)
        d = tmp_path / preset
        loader.save_params({"x": torch.zeros(1)}, str(d), cfg)
        back = loader.config_from_dir(str(d))
        assert back.qkv_bias == cfg.qkv_bias, preset
        assert back.num_experts == cfg.num_experts, preset
        assert back.num_experts_per_tok == cfg.num_experts_per_tok, preset
        assert back.norm_eps == cfg.norm_eps, preset
