"""hipGraph-captured single-token decode.

Batch-1 decode is launch-bound: a 32-layer step is ~300 tiny kernels, and
on MI355X each eager launch costs ~3-4 µs of host time plus ~1.2 µs of
inter-kernel gap (MI355X_MICROARCH.md §boundary / §graph-replay-floor).
Capturing the step in a hipGraph replays the whole token in one ~10-16 µs
submission.

The step is written with static shapes so one capture serves every
position: the position lives in a device tensor (`pos`), KV writes go
through ``index_copy_``, attention runs on the graph-safe flash-decode
kernel (it reads the sequence length from a device int32 at replay time;
the earlier full-window SDPA + repeat_interleave fallback cost a 15 GiB
model ~0.6 ms/token), and the argmax feeds the input buffer back — so N
tokens are N graph replays with no host round-trips.

The same step runs eagerly (CPU or GPU), which is how numerics are tested
against the dynamic-shape forward in models/llama.py.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from fma_amd.ops.decode_ops import (available as _fused_available,
                                    fast_rope_qkv_store,
                                    fast_rmsnorm, fast_rope1,
                                    fast_silu_mul)
from fma_amd.ops.linear import (fast_linear, fast_linear_multi,
                                fast_linear_residual)

from fma_amd.models.llama import LlamaModel, rmsnorm


#: hipGraph replay with vocab sizes beyond this measured boundary hits a
#: hardware exception on ROCm 7.2 (reproduced at vocab 152064 on BOTH the
#: all-custom-kernel and the all-torch-op step, eager always clean;
#: vocab <= 32768 graphs are stable across the whole test matrix and
#: isolated big-vocab argmax/embedding/GEMV micro-graphs pass — the fault
#: needs the full capture pool). Until root-caused, big-vocab models
#: decode eagerly; see NOTES.md round-3 candidates.
GRAPH_SAFE_VOCAB = 100_000


class StaticDecoder:
    """Single-token decode with static shapes over a private KV cache."""

    @staticmethod
    def supported(cfg) -> bool:
        # MoE routing is data-dependent per token: un-capturable;
        # qk_norm (qwen3) is outside the fused qkv+rope capture set
        return (cfg.vocab_size <= GRAPH_SAFE_VOCAB and not cfg.num_experts
                and not cfg.qk_norm)

    def __init__(self, model: LlamaModel, batch: int, max_seq: int):
        assert model.tp_size == 1, "graphed decode is single-rank for now"
        self.model = model
        self.cfg = model.cfg
        self.batch = batch
        self.max_seq = max_seq
        dev = model.device
        cfg = self.cfg
        kv_heads = cfg.num_kv_heads
        self.cache = torch.zeros(
            (cfg.num_layers, 2, batch, max_seq, kv_heads, cfg.head_dim),
            dtype=cfg.dtype, device=dev)
        self.input_tok = torch.zeros((batch, 1), dtype=torch.long, device=dev)
        self.pos = torch.zeros((), dtype=torch.long, device=dev)
        self.positions = torch.arange(max_seq, device=dev)
        self.logits = torch.zeros((batch, cfg.vocab_size), dtype=torch.float32,
                                  device=dev)
        self.out_tokens = torch.zeros((batch, max_seq), dtype=torch.long,
                                      device=dev)
        # runtime sequence length for the graph-safe attention kernel
        # (device int32 read by the kernel itself at replay time)
        self.t_i32 = torch.ones((1,), dtype=torch.int32, device=dev)
        self.pos_i32 = torch.zeros((1,), dtype=torch.int32, device=dev)
        self.graph: Optional[torch.cuda.CUDAGraph] = None

    # -- the static step (graph-capturable) --------------------------------

    def step_(self) -> None:
        """One decode step: reads input_tok/pos, writes cache, logits,
        next token back into input_tok, and out_tokens[:, pos]."""
        m = self.model
        cfg = self.cfg
        P = m.params
        B = self.batch
        q_heads = cfg.num_heads
        kv_heads = cfg.num_kv_heads
        hd = cfg.head_dim

        x = F.embedding(self.input_tok, P["embed.weight"])  # [B,1,H]
        cos = m.rope_cos.index_select(0, self.pos.view(1))  # [1, hd/2]
        sin = m.rope_sin.index_select(0, self.pos.view(1))
        # graph-safe fused attention: the kernel reads t from this device
        # scalar, so the captured launch (sized for max_seq) replays
        # correctly at every position — no full-window SDPA, no
        # repeat_interleave materialization (those cost a 15 GiB model
        # ~0.6 ms/token: 192 -> 173 tok/s measured)
        fused_attn = (B == 1 and x.is_cuda and x.dtype == torch.bfloat16
                      and hd % 64 == 0 and hd <= 256
                      and q_heads % kv_heads == 0 and _fused_available())
        if fused_attn:
            self.t_i32.copy_((self.pos + 1).to(torch.int32))
        self.pos_i32.copy_(self.pos.to(torch.int32).view(1))
        # additive mask over the full window: position j attends iff j <= pos
        neg = -1e9  # large finite: stays finite in bf16
        mask = torch.where(self.positions <= self.pos,
                           torch.zeros((), device=x.device),
                           torch.full((), neg, device=x.device))
        mask = mask.view(1, 1, 1, self.max_seq)

        for li in range(cfg.num_layers):
            p = f"layers.{li}."
            h = fast_rmsnorm(x, P[p + "attn_norm.weight"], cfg.norm_eps)
            q, k, v = fast_linear_multi(
                h, (P[p + "wq.weight"], P[p + "wk.weight"],
                    P[p + "wv.weight"]),
                biases=(P[p + "wq.bias"], P[p + "wk.bias"],
                        P[p + "wv.bias"]) if cfg.qkv_bias else None)
            q = q.view(B, 1, q_heads, hd)
            k = k.view(B, 1, kv_heads, hd)
            v = v.view(B, 1, kv_heads, hd)
            stored = False
            if B == 1 and q.is_cuda and q.dtype == torch.bfloat16:
                # one graph-safe launch: RoPE(q) + RoPE(k)->cache +
                # v->cache, position read from the device scalar at
                # replay time
                stored = fast_rope_qkv_store(
                    q, k, v, self.cache[li, 0, 0], self.cache[li, 1, 0],
                    m.rope_cos, m.rope_sin, pos_dev=self.pos_i32)
            if not stored:
                if q.is_cuda and q.dtype == torch.bfloat16:
                    q = fast_rope1(q, cos[0], sin[0])
                    k = fast_rope1(k, cos[0], sin[0])
                else:
                    q = _rope1(q, cos, sin)
                    k = _rope1(k, cos, sin)
                # static cache write at pos
                self.cache[li, 0].index_copy_(1, self.pos.view(1), k)
                self.cache[li, 1].index_copy_(1, self.pos.view(1), v)
            if fused_attn:
                from fma_amd.ops import actuation
                att = actuation._C.attn_decode_bf16_graph(
                    q.reshape(q_heads, hd).contiguous(),
                    self.cache[li, 0, 0], self.cache[li, 1, 0],
                    self.t_i32, self.max_seq)
                att = att.view(B, 1, q_heads * hd)
            else:
                kh = self.cache[li, 0].transpose(1, 2)  # [B, kvH, S, hd]
                vh = self.cache[li, 1].transpose(1, 2)
                if kv_heads != q_heads:
                    rep = q_heads // kv_heads
                    kh = kh.repeat_interleave(rep, dim=1)
                    vh = vh.repeat_interleave(rep, dim=1)
                att = F.scaled_dot_product_attention(
                    q.transpose(1, 2), kh, vh, attn_mask=mask.to(q.dtype))
                att = att.transpose(1, 2).reshape(B, 1, q_heads * hd)
            if x.is_cuda and x.dtype == torch.bfloat16 \
                    and x.numel() == x.shape[-1]:
                x = fast_linear_residual(att, P[p + "wo.weight"], x)
                h = fast_rmsnorm(x, P[p + "mlp_norm.weight"], cfg.norm_eps)
                gate, up = fast_linear_multi(
                    h, (P[p + "w_gate.weight"], P[p + "w_up.weight"]))
                x = fast_linear_residual(fast_silu_mul(gate, up),
                                         P[p + "w_down.weight"], x)
            else:
                x = x + fast_linear(att, P[p + "wo.weight"])
                h = fast_rmsnorm(x, P[p + "mlp_norm.weight"], cfg.norm_eps)
                gate = fast_linear(h, P[p + "w_gate.weight"])
                up = fast_linear(h, P[p + "w_up.weight"])
                x = x + fast_linear(fast_silu_mul(gate, up),
                                    P[p + "w_down.weight"])

        x = fast_rmsnorm(x, P["final_norm.weight"], cfg.norm_eps)
        logits = fast_linear(x[:, 0], P["lm_head.weight"]).float()
        self.logits.copy_(logits)
        nxt = logits.argmax(-1, keepdim=True)
        self.pos.add_(1)
        self.out_tokens.index_copy_(1, self.pos.view(1).clamp(
            max=self.max_seq - 1), nxt)
        self.input_tok.copy_(nxt)

    # -- capture / replay ----------------------------------------------------

    def capture(self) -> None:
        """Record the step as a hipGraph (GPU only). Warm up twice on a side
        stream first (allocator + RCCL-free step), then capture."""
        assert self.model.device.type == "cuda"
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self.step_()
        torch.cuda.current_stream().wait_stream(s)
        # reset state disturbed by warmup
        self.pos.zero_()
        self.cache.zero_()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self.step_()
        self.graph = g
        self.pos.zero_()
        self.cache.zero_()

    @torch.no_grad()
    def prefill(self, prompt: torch.Tensor) -> None:
        """Batched eager prefill into the decoder's cache; ends in the same
        state the step loop would reach (pos=T, first prediction staged)."""
        from types import SimpleNamespace

        B, T = prompt.shape
        assert B == self.batch and T < self.max_seq
        self.pos.zero_()
        self.cache.zero_()
        self.out_tokens.zero_()
        if T > 1:
            cache_view = SimpleNamespace(data=self.cache)
            logits = self.model.forward(prompt, cache_view, 0)
            nxt = logits[:, -1].argmax(-1, keepdim=True)
            self.pos.fill_(T)
            self.out_tokens[:, T:T + 1] = nxt
            self.input_tok.copy_(nxt)
            self.logits.copy_(logits[:, -1])
        else:
            self.input_tok.copy_(prompt[:, 0:1])
            self._one()

    def _one(self) -> None:
        if self.graph is not None:
            self.graph.replay()
        else:
            self.step_()

    @torch.no_grad()
    def generate(self, prompt: torch.Tensor, max_new_tokens: int
                 ) -> torch.Tensor:
        """Greedy decode; returns [B, T + max_new_tokens] like
        LlamaModel.generate."""
        self.prefill(prompt)
        # after prefill, input_tok holds the first generated token
        for _ in range(max_new_tokens - 1):
            self._one()
        T = prompt.shape[1]
        gen = self.out_tokens[:, T:T + max_new_tokens]
        return torch.cat([prompt, gen], dim=1)


def _rope1(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor
           ) -> torch.Tensor:
    # x: [B,1,H,hd]; cos/sin: [1, hd/2]
    B, T, H, D = x.shape
    xf = x.float().view(B, T, H, D // 2, 2)
    c = cos.view(1, T, 1, D // 2)
    s = sin.view(1, T, 1, D // 2)
    x0, x1 = xf[..., 0], xf[..., 1]
    out = torch.stack((x0 * c - x1 * s, x0 * s + x1 * c), dim=-1)
    return out.view(B, T, H, D).to(x.dtype)
