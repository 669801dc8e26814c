"""Static decoder numerics vs the dynamic-shape forward (CPU)."""

import os

import pytest
import torch

os.environ.setdefault("FMA_FAKE_GPU", "1")

from fma_amd.models.decode_graph import StaticDecoder  # noqa: E402
from fma_amd.models.llama import LlamaConfig  # noqa: E402
from fma_amd.runtime.engine import ActuationEngine  # noqa: E402


@pytest.fixture(scope="module")
def engine():
    return ActuationEngine(LlamaConfig.tiny(), seed=13)


def test_static_decode_matches_eager_generate(engine):
    torch.manual_seed(0)
    prompt = torch.randint(0, engine.cfg.vocab_size, (1, 7))
    eager = engine.model.generate(prompt, max_new_tokens=5)
    dec = StaticDecoder(engine.model, batch=1, max_seq=32)
    static = dec.generate(prompt, max_new_tokens=5)
    assert torch.equal(eager, static), (eager, static)


def test_static_decoder_reusable_across_prompts(engine):
    dec = StaticDecoder(engine.model, batch=1, max_seq=32)
    for seed in (1, 2):
        torch.manual_seed(seed)
        prompt = torch.randint(0, engine.cfg.vocab_size, (1, 4))
        eager = engine.model.generate(prompt, max_new_tokens=3)
        assert torch.equal(dec.generate(prompt, max_new_tokens=3), eager)


def test_static_decoder_logits_match_forward(engine):
    torch.manual_seed(3)
    prompt = torch.randint(0, engine.cfg.vocab_size, (1, 5))
    dec = StaticDecoder(engine.model, batch=1, max_seq=16)
    dec.prefill(prompt)
    full = engine.model.forward(prompt)  # [1, 5, V]
    # bf16 accumulation-order differences (padded-window SDPA vs exact
    # length) bound the agreement; greedy decode is exact (tests above)
    assert torch.allclose(dec.logits, full[:, -1], atol=1e-2, rtol=1e-2)


def test_graph_decoder_gated_on_safe_vocab():
    """Big-vocab models must not select the hipGraph decoder (measured
    ROCm 7.2 replay fault at vocab 152064; decode_graph.GRAPH_SAFE_VOCAB
    documents the boundary). Eager decode remains the path."""
    from fma_amd.models.decode_graph import GRAPH_SAFE_VOCAB, StaticDecoder
    from fma_amd.models.llama import LlamaConfig

    assert StaticDecoder.supported(LlamaConfig.by_name("synthetic-15gib"))
    qwen = LlamaConfig.by_name("qwen2-7b")
    assert qwen.vocab_size > GRAPH_SAFE_VOCAB
    assert not StaticDecoder.supported(qwen)
