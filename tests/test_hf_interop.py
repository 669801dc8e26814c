"""HuggingFace checkpoint interop, cross-validated against transformers.

A user of the reference serves HF safetensors models through vLLM
(reference docs/dual-pods.md:599-608); here the loader maps transformers
naming onto our parameter schema (loader.map_hf_name), fixes up the RoPE
layout difference (rotate-half vs adjacent-pair, loader._unrotate_half)
and ties lm_head when the checkpoint omits it. These tests are the
strongest correctness statement in the repo's CPU tier: random-init
transformers models are saved with save_pretrained and our model must
reproduce their LOGITS — validating the converter AND our entire model
math (rmsnorm, rope, GQA attention, silu MLP, MoE routing) against the
canonical implementation.
"""

import pytest
import torch

from fma_amd.models import loader
from fma_amd.models.llama import LlamaModel

transformers = pytest.importorskip("transformers")

pytestmark = pytest.mark.timeout(240)


def _load_ours(ckpt_dir):
    cfg = loader.config_from_dir(ckpt_dir)
    cfg.dtype = torch.float32  # tight comparison against fp32 HF
    params = {n: torch.zeros(s, dtype=d)
              for n, s, d in cfg.param_specs()}
    loader.load_into_params(ckpt_dir, params, cfg=cfg)
    return cfg, LlamaModel(cfg, params, torch.device("cpu"))


def _compare_logits(hf_model, ckpt_dir, rtol=2e-3, atol=2e-3):
    cfg, ours = _load_ours(ckpt_dir)
    torch.manual_seed(7)
    tokens = torch.randint(0, cfg.vocab_size, (1, 12))
    with torch.no_grad():
        ref = hf_model(tokens).logits
        got = ours.forward(tokens)
    assert torch.allclose(got, ref, rtol=rtol, atol=atol), \
        (got - ref).abs().max().item()


def test_llama_hf_checkpoint_logits_match(tmp_path):
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, rope_theta=10000.0, rms_norm_eps=1e-5,
        tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    _compare_logits(hf, str(tmp_path))


def test_llama_hf_untied_head(tmp_path):
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=96, hidden_size=32, intermediate_size=48,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=1,
        max_position_embeddings=32, tie_word_embeddings=False)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    _compare_logits(hf, str(tmp_path))


def test_qwen2_hf_checkpoint_logits_match(tmp_path):
    """Qwen2: attention-projection biases + qkv_bias inferred from
    model_type in the HF config.json."""
    from transformers import Qwen2Config, Qwen2ForCausalLM
    hf = Qwen2ForCausalLM(Qwen2Config(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, rope_theta=10000.0,
        tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    assert cfg.qkv_bias  # inferred from model_type == "qwen2"
    _compare_logits(hf, str(tmp_path))


def test_mixtral_hf_checkpoint_logits_match(tmp_path):
    """Mixtral: router (gate) + experts w1/w3/w2 naming, top-2 routing
    with post-topk softmax."""
    from transformers import MixtralConfig, MixtralForCausalLM
    hf = MixtralForCausalLM(MixtralConfig(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2,
        max_position_embeddings=64, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    assert cfg.num_experts == 4 and cfg.num_experts_per_tok == 2
    _compare_logits(hf, str(tmp_path))


def test_map_hf_name_table():
    cases = {
        "model.embed_tokens.weight": "embed.weight",
        "model.norm.weight": "final_norm.weight",
        "lm_head.weight": "lm_head.weight",
        "model.layers.3.self_attn.q_proj.weight": "layers.3.wq.weight",
        "model.layers.3.self_attn.k_proj.bias": "layers.3.wk.bias",
        "model.layers.0.input_layernorm.weight":
            "layers.0.attn_norm.weight",
        "model.layers.0.post_attention_layernorm.weight":
            "layers.0.mlp_norm.weight",
        "model.layers.1.mlp.gate_proj.weight": "layers.1.w_gate.weight",
        "model.layers.1.mlp.down_proj.weight": "layers.1.w_down.weight",
        "model.layers.2.block_sparse_moe.gate.weight":
            "layers.2.router.weight",
        "model.layers.2.block_sparse_moe.experts.5.w1.weight":
            "layers.2.experts.5.w_gate.weight",
        "model.layers.2.block_sparse_moe.experts.5.w2.weight":
            "layers.2.experts.5.w_down.weight",
        "model.layers.2.block_sparse_moe.experts.5.w3.weight":
            "layers.2.experts.5.w_up.weight",
        "model.layers.0.self_attn.rotary_emb.inv_freq": None,
        "something.else": None,
    }
    for hf_name, ours in cases.items():
        assert loader.map_hf_name(hf_name) == ours, hf_name


def test_llama_hf_greedy_generation_matches(tmp_path):
    """Token-level equality: greedy decode through OUR KV-cache path vs
    transformers generate on the same HF checkpoint — validates the
    cached incremental forward (rope offsets, cache indexing) end to
    end, not just one prefill."""
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    _, ours = _load_ours(str(tmp_path))
    torch.manual_seed(11)
    prompt = torch.randint(0, 128, (1, 8))
    with torch.no_grad():
        ref = hf.generate(prompt, max_new_tokens=12, do_sample=False,
                          use_cache=True)
        got = ours.generate(prompt, max_new_tokens=12)
    assert torch.equal(got, ref[:, :got.shape[1]]), (got, ref)


def test_hf_checkpoint_tp_sharding_composes(tmp_path):
    """The HF rope permutation is per-head, so it must commute with
    Megatron head-aligned sharding: TP shards of an HF checkpoint glued
    back together equal the TP=1 load — including the KV-replication
    branch (tp=4 > kv_heads=2)."""
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        num_hidden_layers=1, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=32, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    cfg.dtype = torch.float32
    full = {n: torch.zeros(s, dtype=d) for n, s, d in cfg.param_specs()}
    loader.load_into_params(str(tmp_path), full, cfg=cfg)

    for tp in (2, 4):  # tp=4 exercises KV replication (kv_heads=2)
        shards = []
        for r in range(tp):
            p = {n: torch.zeros(s, dtype=d)
                 for n, s, d in cfg.param_specs(r, tp)}
            loader.load_into_params(str(tmp_path), p, tp_rank=r,
                                    tp_size=tp, cfg=cfg)
            shards.append(p)
        name = "layers.0.wq.weight"
        glued = torch.cat([s[name] for s in shards], dim=0)
        assert torch.equal(glued, full[name]), f"tp={tp} wq"
        name = "layers.0.wo.weight"
        glued = torch.cat([s[name] for s in shards], dim=1)
        assert torch.equal(glued, full[name]), f"tp={tp} wo"
        # wk: sliced at tp=2, replicated by KV group at tp=4
        name = "layers.0.wk.weight"
        kvh, hd = cfg.num_kv_heads, cfg.head_dim
        for r in range(tp):
            if tp <= kvh:
                lo = r * (kvh // tp) * hd
                want = full[name][lo:lo + (kvh // tp) * hd]
            else:
                g = r * kvh // tp
                want = full[name][g * hd:(g + 1) * hd]
            assert torch.equal(shards[r][name], want), f"tp={tp} r={r} wk"


def test_engine_arena_fast_path_loads_hf(tmp_path):
    """The engine's pinned fast-load path (stage -> host buffer -> arena)
    converts HF tensors too, including mirroring embed bytes into the
    tied lm_head slot; greedy tokens match transformers."""
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM

    from fma_amd.runtime.engine import ActuationEngine

    hf = LlamaForCausalLM(HFConfig(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    eng = ActuationEngine(cfg, init_weights=False)
    assert eng.actuation_mode == "arena" and eng.arena is not None
    eng.load_checkpoint(str(tmp_path))
    assert torch.equal(eng.params["lm_head.weight"],
                       eng.params["embed.weight"])  # tied
    torch.manual_seed(5)
    prompt = torch.randint(0, 128, (1, 6))
    with torch.no_grad():
        ref = hf.generate(prompt, max_new_tokens=4, do_sample=False,
                          use_cache=True)
    got = eng.generate(prompt, max_new_tokens=4)
    # engine runs bf16: agreement on prompt + first generated tokens
    assert torch.equal(got[0, :8], ref[0, :8])


def test_llama_hf_explicit_head_dim(tmp_path):
    """Models whose head_dim differs from hidden//heads (HF configs
    carry an explicit "head_dim") load and match transformers."""
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=2,
        head_dim=16,  # derived would be 32
        max_position_embeddings=32, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    assert cfg.head_dim == 16
    _compare_logits(hf, str(tmp_path))


def test_llama31_rope_scaling_matches(tmp_path):
    """llama-3.1-style rope_scaling ("llama3" type): scaled frequencies
    must reproduce transformers' logits — plain rope would diverge."""
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=2,
        max_position_embeddings=64, rope_theta=10000.0,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 16},
        tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    assert cfg.rope_scaling and cfg.rope_scaling["factor"] == 8.0
    _compare_logits(hf, str(tmp_path))
    # sanity: ignoring the scaling WOULD diverge (the test has teeth)
    cfg2 = loader.config_from_dir(str(tmp_path))
    cfg2.dtype = torch.float32
    cfg2.rope_scaling = None
    params = {n: torch.zeros(s, dtype=d)
              for n, s, d in cfg2.param_specs()}
    loader.load_into_params(str(tmp_path), params, cfg=cfg2)
    unscaled = LlamaModel(cfg2, params, torch.device("cpu"))
    torch.manual_seed(7)
    tokens = torch.randint(0, cfg2.vocab_size, (1, 12))
    with torch.no_grad():
        ref = hf(tokens).logits
        got = unscaled.forward(tokens)
    assert not torch.allclose(got, ref, rtol=2e-3, atol=2e-3)


def test_linear_rope_scaling_matches(tmp_path):
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=1,
        max_position_embeddings=64,
        rope_scaling={"rope_type": "linear", "factor": 2.0},
        tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    _compare_logits(hf, str(tmp_path))


def test_sliding_window_checkpoint_refused(tmp_path):
    """A config that actually relies on sliding-window attention is
    refused loudly instead of silently serving wrong numerics."""
    import json
    import os
    os.makedirs(tmp_path, exist_ok=True)
    with open(tmp_path / "config.json", "w") as f:
        json.dump({"model_type": "qwen2", "hidden_size": 64,
                   "num_attention_heads": 2, "num_hidden_layers": 1,
                   "use_sliding_window": True, "sliding_window": 1024,
                   "max_position_embeddings": 32768}, f)
    with pytest.raises(ValueError, match="sliding-window"):
        loader.config_from_dir(str(tmp_path))
    # window >= context: plain full attention is exact; accepted
    with open(tmp_path / "config.json", "w") as f:
        json.dump({"model_type": "qwen2", "hidden_size": 64,
                   "num_attention_heads": 2, "num_hidden_layers": 1,
                   "use_sliding_window": True, "sliding_window": 32768,
                   "max_position_embeddings": 4096}, f)
    assert loader.config_from_dir(str(tmp_path)) is not None


def test_real_tokenizer_from_checkpoint_dir(tmp_path):
    """A tokenizer.json beside the weights makes /v1/completions speak
    real text: encode through the checkpoint's tokenizer, decode the
    generated ids back."""
    pytest.importorskip("tokenizers")
    from tokenizers import Tokenizer
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace

    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    from fma_amd.runtime.server import ServingRuntime, parse_options

    hf = LlamaForCausalLM(HFConfig(
        vocab_size=32, hidden_size=32, intermediate_size=48,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=1,
        max_position_embeddings=32, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    vocab = {f"w{i}": i for i in range(30)}
    vocab["[UNK]"] = 30
    tok = Tokenizer(WordLevel(vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = Whitespace()
    tok.save(str(tmp_path / "tokenizer.json"))

    rt = ServingRuntime(parse_options(f"--model {tmp_path}"))
    eng = rt.rt.engine if hasattr(rt.rt, "engine") else rt.rt
    assert eng.tokenizer is not None
    assert eng.tokenizer.encode("w1 w2 w7").ids == [1, 2, 7]
    text = rt.rt.generate_text("w1 w2 w7", max_new_tokens=4)
    # decoded ids are words from the vocab, not raw bytes
    assert all(t in vocab for t in text.split()), text
    if hasattr(rt.rt, "stop"):
        rt.rt.stop()


def test_hf_mixtral_expert_parallel_slicing(tmp_path):
    """EP over an HF Mixtral checkpoint: each rank loads only its WHOLE
    experts (full tensors, global indices), skipping foreign ones."""
    from transformers import MixtralConfig, MixtralForCausalLM
    hf = MixtralForCausalLM(MixtralConfig(
        vocab_size=64, hidden_size=32, intermediate_size=48,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2,
        max_position_embeddings=32, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    cfg.dtype = torch.float32
    cfg.expert_parallel = True
    for rank in (0, 1):
        params = {n: torch.zeros(s, dtype=d)
                  for n, s, d in cfg.param_specs(rank, 2)}
        mine = list(cfg.local_experts(rank, 2))
        assert mine == ([0, 1] if rank == 0 else [2, 3])
        n = loader.load_into_params(
            str(tmp_path), params, tp_rank=rank, tp_size=2, cfg=cfg,
            skip=lambda name: ".experts." in name)
        assert n == len(params)
        from safetensors import safe_open
        with safe_open(str(tmp_path / "model.safetensors"),
                       framework="pt") as sf:
            for e in mine:
                hf_w1 = sf.get_tensor(
                    f"model.layers.0.block_sparse_moe.experts.{e}"
                    ".w1.weight")
                assert torch.equal(
                    params[f"layers.0.experts.{e}.w_gate.weight"], hf_w1)


def test_qwen3_hf_checkpoint_logits_match(tmp_path):
    """Qwen3: per-head q/k RMSNorm before rope (the norm gain permutes
    like a single head's projection rows), explicit head_dim, no
    attention biases."""
    from transformers import Qwen3Config, Qwen3ForCausalLM
    hf = Qwen3ForCausalLM(Qwen3Config(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=64,
        tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    assert cfg.qk_norm and not cfg.qkv_bias and cfg.head_dim == 16
    _compare_logits(hf, str(tmp_path))
    # greedy tokens incl. the cached decode path
    _, ours = _load_ours(str(tmp_path))
    torch.manual_seed(3)
    prompt = torch.randint(0, 96, (1, 7))
    with torch.no_grad():
        ref = hf.generate(prompt, max_new_tokens=8, do_sample=False,
                          use_cache=True)
        got = ours.generate(prompt, max_new_tokens=8)
    assert torch.equal(got, ref[:, :got.shape[1]])


def test_gemma_hf_checkpoint_logits_match(tmp_path):
    """Gemma: gelu-tanh gated MLP, sqrt(hidden) embedding scale, and the
    (1+w) rmsnorm offset folded into the gains at load — logits match
    transformers."""
    from transformers import GemmaConfig as HFGemma, GemmaForCausalLM
    hf = GemmaForCausalLM(HFGemma(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=64,
        tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    assert cfg.hidden_act == "gelu_tanh" and cfg.embed_scale
    _compare_logits(hf, str(tmp_path))
    _, ours = _load_ours(str(tmp_path))
    torch.manual_seed(3)
    prompt = torch.randint(0, 96, (1, 7))
    with torch.no_grad():
        ref = hf.generate(prompt, max_new_tokens=8, do_sample=False,
                          use_cache=True)
        got = ours.generate(prompt, max_new_tokens=8)
    assert torch.equal(got, ref[:, :got.shape[1]])


def test_mistral_hf_checkpoint_logits_match(tmp_path):
    """Mistral (v0.3-style, no sliding window): llama-architecture with
    its own model_type — loads and matches transformers."""
    from transformers import MistralConfig, MistralForCausalLM
    hf = MistralForCausalLM(MistralConfig(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        sliding_window=None, max_position_embeddings=64,
        tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    _compare_logits(hf, str(tmp_path))


def test_sharded_hf_checkpoint_loads(tmp_path):
    """Multi-file HF checkpoints (model-0000N-of-0000M.safetensors +
    index) load the same as single-file ones."""
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True,
                       max_shard_size="50KB")
    import glob as _g
    shards = _g.glob(str(tmp_path / "model-*.safetensors"))
    assert len(shards) >= 2, "expected a sharded save"
    _compare_logits(hf, str(tmp_path))


def test_generate_stops_at_eos(tmp_path):
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=64, hidden_size=32, intermediate_size=48,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=1,
        max_position_embeddings=32, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    _, ours = _load_ours(str(tmp_path))
    torch.manual_seed(2)
    prompt = torch.randint(0, 64, (1, 4))
    free_run = ours.generate(prompt, max_new_tokens=8)
    first = int(free_run[0, 4])
    stopped = ours.generate(prompt, max_new_tokens=8, eos_id=first)
    assert stopped.shape[1] == 5 and int(stopped[0, -1]) == first


@pytest.mark.parametrize("norm_topk", [False, True])
def test_qwen3_moe_hf_checkpoint_logits_match(tmp_path, norm_topk):
    """Qwen3-MoE: qk norms + sparse MoE with its own expert width
    (moe_intermediate_size) and qwen3-style router naming; both
    norm_topk_prob semantics (global-softmax top-k vs renormalized)."""
    from transformers import Qwen3MoeConfig, Qwen3MoeForCausalLM
    hf = Qwen3MoeForCausalLM(Qwen3MoeConfig(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        moe_intermediate_size=48, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, head_dim=16,
        num_experts=4, num_experts_per_tok=2, norm_topk_prob=norm_topk,
        max_position_embeddings=64, tie_word_embeddings=True)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    cfg = loader.config_from_dir(str(tmp_path))
    assert cfg.qk_norm and cfg.num_experts == 4
    assert cfg.moe_intermediate_size == 48
    assert cfg.moe_norm_topk is norm_topk
    _compare_logits(hf, str(tmp_path))


def test_phi3_hf_checkpoint_logits_match(tmp_path):
    """Phi-3: FUSED qkv_proj / gate_up_proj tensors split at load (with
    the rope row-permutation applied to the q/k slices)."""
    from transformers import Phi3Config, Phi3ForCausalLM
    hf = Phi3ForCausalLM(Phi3Config(
        vocab_size=96, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, eos_token_id=2, pad_token_id=0,
        bos_token_id=1, tie_word_embeddings=False)).eval()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    _compare_logits(hf, str(tmp_path))
    # arena fast path splits the fused tensors too
    from fma_amd.runtime.engine import ActuationEngine
    cfg = loader.config_from_dir(str(tmp_path))
    eng = ActuationEngine(cfg, init_weights=False)
    eng.load_checkpoint(str(tmp_path))
    torch.manual_seed(5)
    prompt = torch.randint(0, 96, (1, 6))
    with torch.no_grad():
        ref = hf.generate(prompt, max_new_tokens=4, do_sample=False,
                          use_cache=True)
    got = eng.generate(prompt, max_new_tokens=4)
    assert torch.equal(got[0, :8], ref[0, :8])


def test_hf_check_tool(tmp_path, capsys):
    """tools/hf_check.py: the operator's will-this-serve probe."""
    import sys
    from transformers import LlamaConfig as HFConfig, LlamaForCausalLM
    hf = LlamaForCausalLM(HFConfig(
        vocab_size=64, hidden_size=32, intermediate_size=48,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=1,
        max_position_embeddings=32, tie_word_embeddings=True))
    hf.save_pretrained(tmp_path, safe_serialization=True)
    sys.path.insert(0, "tools")
    import hf_check
    old = sys.argv
    sys.argv = ["hf_check", str(tmp_path), "--load"]
    try:
        assert hf_check.main() == 0
    finally:
        sys.argv = old
    out = capsys.readouterr().out
    assert "load OK" in out and "finite=True" in out
