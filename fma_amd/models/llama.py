"""Llama-architecture model with arena-resident weights.

The serving runtime materializes every parameter as a view into one
contiguous :class:`~fma_amd.ops.actuation.ArenaActuator` region, which is
what makes sleep(level=1)/wake_up a pure pinned-transfer (no per-tensor
gather needed on the hot path). Weights are random-initialized — this stack
has no network access; BASELINE.md's configs all use random-init weights of
the named architecture — or loaded from safetensors when a file is present.

Tensor-parallel layout (Megatron-style, one process per GPU over RCCL):
q/k/v and gate/up are row-sharded (output features / tp), o and down are
column-sharded (input features / tp) with one all-reduce each per layer;
embeddings, norms and lm_head are replicated. The reference has no model
code at all — parallelism there is an opaque ``--tensor-parallel-size``
passed to vLLM (reference docs/launcher.md:584-595); here it is native.

Inference only: plain tensors, no autograd.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn.functional as F

from fma_amd.ops.decode_ops import (fast_attn_decode, fast_attn_prefill,
                                    fast_rmsnorm, fast_rope1,
                                    fast_rope_qkv_store, fast_silu_mul)
from fma_amd.ops.linear import (fast_linear, fast_linear_multi,
                                fast_linear_residual)


@dataclass
class LlamaConfig:
    name: str = "custom"
    vocab_size: int = 32768
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    max_seq_len: int = 4096
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    #: Qwen2-family: biases on the q/k/v projections (the only
    #: architectural delta vs llama the actuation path sees)
    qkv_bias: bool = False
    #: Qwen3-family: RMSNorm over head_dim on q and k before rope
    #: (one (hd,) gain shared across heads, per layer)
    qk_norm: bool = False
    #: MLP activation: "silu" (llama/qwen/mixtral) or "gelu_tanh"
    #: (gemma). The fused silu kernel only engages for silu.
    hidden_act: str = "silu"
    #: Gemma: hidden states scale by sqrt(hidden_size) after embedding.
    #: (Gemma's (1+w) rmsnorm offset is folded into the weights at
    #: checkpoint load — loader.hf_convert — so the runtime norm is
    #: the standard one.)
    embed_scale: bool = False
    #: Mixtral-family: sparse MoE MLP (0 = dense). Routed top-k over
    #: num_experts per token; each expert is a llama-shaped gated MLP.
    num_experts: int = 0
    num_experts_per_tok: int = 2
    #: expert FFN width when it differs from intermediate_size
    #: (qwen3-moe's moe_intermediate_size); None = intermediate_size
    moe_intermediate_size: "int | None" = None
    #: router weight normalization: True = softmax over the selected
    #: top-k (mixtral; equals qwen3-moe norm_topk_prob=true), False =
    #: softmax over ALL experts, top-k taken unrenormalized
    #: (qwen3-moe norm_topk_prob=false)
    moe_norm_topk: bool = True
    #: EP: partition WHOLE experts across ranks (attention stays
    #: Megatron-TP); non-owners contribute zero and the layer's single
    #: all-reduce sums the routed outputs. False = Megatron-shard every
    #: expert's MLP like the dense path.
    expert_parallel: bool = False
    #: explicit per-head dim for models where it differs from
    #: hidden_size // num_heads (HF configs carry "head_dim"); None
    #: derives it
    head_dim_override: "int | None" = None
    #: HF-style rope_scaling dict ({"rope_type": "llama3"|"linear",
    #: "factor": ..., ...}) — llama-3.1+ checkpoints need it for
    #: correct frequencies; None = plain rope
    rope_scaling: "dict | None" = None
    dtype: torch.dtype = torch.bfloat16

    @property
    def head_dim(self) -> int:
        return self.head_dim_override or self.hidden_size // self.num_heads

    def local_experts(self, tp_rank: int = 0, tp_size: int = 1
                      ) -> range:
        """Global indices of the experts this rank holds."""
        if not self.num_experts:
            return range(0)
        if not (self.expert_parallel and tp_size > 1):
            return range(self.num_experts)
        assert self.num_experts % tp_size == 0, \
            "EP needs num_experts divisible by tp_size"
        per = self.num_experts // tp_size
        return range(tp_rank * per, (tp_rank + 1) * per)

    # -- presets ------------------------------------------------------------

    @staticmethod
    def tiny() -> "LlamaConfig":
        return LlamaConfig(name="tiny", vocab_size=512, hidden_size=64,
                           intermediate_size=128, num_layers=2, num_heads=4,
                           num_kv_heads=2, max_seq_len=256)

    @staticmethod
    def qwen2_7b() -> "LlamaConfig":
        return LlamaConfig(name="qwen2-7b", vocab_size=152064,
                           hidden_size=3584, intermediate_size=18944,
                           num_layers=28, num_heads=28, num_kv_heads=4,
                           max_seq_len=8192, rope_theta=1000000.0,
                           norm_eps=1e-6, qkv_bias=True)

    @staticmethod
    def tiny_qwen() -> "LlamaConfig":
        return LlamaConfig(name="tiny-qwen", vocab_size=512, hidden_size=64,
                           intermediate_size=128, num_layers=2, num_heads=4,
                           num_kv_heads=2, max_seq_len=256, qkv_bias=True)

    @staticmethod
    def tiny_qwen3() -> "LlamaConfig":
        return LlamaConfig(name="tiny-qwen3", vocab_size=512,
                           hidden_size=64, intermediate_size=128,
                           num_layers=2, num_heads=4, num_kv_heads=2,
                           max_seq_len=256, qk_norm=True)

    @staticmethod
    def qwen3_8b() -> "LlamaConfig":
        # Qwen3-8B shape: explicit head_dim 128, GQA 32/8, qk norms
        return LlamaConfig(name="qwen3-8b", vocab_size=151936,
                           hidden_size=4096, intermediate_size=12288,
                           num_layers=36, num_heads=32, num_kv_heads=8,
                           head_dim_override=128, max_seq_len=8192,
                           rope_theta=1000000.0, qk_norm=True)

    @staticmethod
    def tiny_gemma() -> "LlamaConfig":
        return LlamaConfig(name="tiny-gemma", vocab_size=512,
                           hidden_size=64, intermediate_size=128,
                           num_layers=2, num_heads=4, num_kv_heads=2,
                           head_dim_override=16, max_seq_len=256,
                           hidden_act="gelu_tanh", embed_scale=True)

    @staticmethod
    def mixtral_8x7b() -> "LlamaConfig":
        return LlamaConfig(name="mixtral-8x7b", vocab_size=32000,
                           hidden_size=4096, intermediate_size=14336,
                           num_layers=32, num_heads=32, num_kv_heads=8,
                           max_seq_len=8192, rope_theta=1000000.0,
                           num_experts=8, num_experts_per_tok=2)

    @staticmethod
    def tiny_moe() -> "LlamaConfig":
        return LlamaConfig(name="tiny-moe", vocab_size=512, hidden_size=64,
                           intermediate_size=96, num_layers=2, num_heads=4,
                           num_kv_heads=2, max_seq_len=256, num_experts=4,
                           num_experts_per_tok=2)

    @staticmethod
    def llama3_8b() -> "LlamaConfig":
        return LlamaConfig(name="llama-3-8b", vocab_size=128256,
                           hidden_size=4096, intermediate_size=14336,
                           num_layers=32, num_heads=32, num_kv_heads=8,
                           max_seq_len=8192)

    @staticmethod
    def llama3_70b() -> "LlamaConfig":
        return LlamaConfig(name="llama-3-70b", vocab_size=128256,
                           hidden_size=8192, intermediate_size=28672,
                           num_layers=80, num_heads=64, num_kv_heads=8,
                           max_seq_len=8192)

    @staticmethod
    def from_total_gib(gib: float, dtype: torch.dtype = torch.bfloat16
                       ) -> "LlamaConfig":
        """Synthetic 70B-shaped config whose total parameter bytes are
        ~`gib` GiB — used for BASELINE's '64 GiB of tensors' wake metric.
        Small targets shrink the base dims so layer count stays sane."""
        if gib >= 16:
            dims = dict(vocab_size=32768, hidden_size=8192,
                        intermediate_size=28672, num_heads=64, num_kv_heads=8)
        elif gib >= 2:
            dims = dict(vocab_size=32768, hidden_size=4096,
                        intermediate_size=14336, num_heads=32, num_kv_heads=8)
        else:
            dims = dict(vocab_size=4096, hidden_size=1024,
                        intermediate_size=3584, num_heads=8, num_kv_heads=8)
        cfg = LlamaConfig(name=f"synthetic-{gib:g}gib", num_layers=1,
                          max_seq_len=4096, dtype=dtype, **dims)
        esize = torch.empty(0, dtype=dtype).element_size()
        fixed = (2 * cfg.vocab_size * cfg.hidden_size + cfg.hidden_size) * esize
        kv_dim = cfg.num_kv_heads * cfg.head_dim
        per_layer = esize * (
            2 * cfg.hidden_size * cfg.hidden_size            # wq, wo
            + 2 * kv_dim * cfg.hidden_size                   # wk, wv
            + 3 * cfg.hidden_size * cfg.intermediate_size    # gate, up, down
            + 2 * cfg.hidden_size)                           # norms
        target = gib * (1 << 30)
        cfg.num_layers = max(1, round((target - fixed) / per_layer))
        return cfg

    @staticmethod
    def by_name(name: str) -> "LlamaConfig":
        presets = {
            "tiny": LlamaConfig.tiny,
            "tiny-qwen": LlamaConfig.tiny_qwen,
            "tiny-moe": LlamaConfig.tiny_moe,
            "tiny-qwen3": LlamaConfig.tiny_qwen3,
            "tiny-gemma": LlamaConfig.tiny_gemma,
            "qwen3-8b": LlamaConfig.qwen3_8b,
            "llama-3-8b": LlamaConfig.llama3_8b,
            "llama-3-70b": LlamaConfig.llama3_70b,
            "qwen2-7b": LlamaConfig.qwen2_7b,
            "mixtral-8x7b": LlamaConfig.mixtral_8x7b,
        }
        if name in presets:
            return presets[name]()
        if name.startswith("synthetic-") and name.endswith("gib"):
            return LlamaConfig.from_total_gib(float(name[len("synthetic-"):-3]))
        raise KeyError(f"unknown model preset {name!r}")

    # -- parameter schema ---------------------------------------------------

    def param_specs(self, tp_rank: int = 0, tp_size: int = 1
                    ) -> List[Tuple[str, Tuple[int, ...], torch.dtype]]:
        """Names/shapes of this rank's parameter shard, in layout order."""
        assert self.num_heads % tp_size == 0, "heads must divide tp"
        assert (self.num_kv_heads % tp_size == 0
                or tp_size % self.num_kv_heads == 0), \
            "tp must divide kv heads or be a multiple (KV replication)"
        h = self.hidden_size
        hd = self.head_dim
        q_local = self.num_heads // tp_size * hd
        kv_local = max(self.num_kv_heads // tp_size, 1) * hd
        i_local = self.intermediate_size // tp_size
        d = self.dtype
        specs: List[Tuple[str, Tuple[int, ...], torch.dtype]] = [
            ("embed.weight", (self.vocab_size, h), d),
        ]
        for li in range(self.num_layers):
            p = f"layers.{li}."
            specs += [
                (p + "attn_norm.weight", (h,), d),
                (p + "wq.weight", (q_local, h), d),
                (p + "wk.weight", (kv_local, h), d),
                (p + "wv.weight", (kv_local, h), d),
            ]
            if self.qkv_bias:
                specs += [
                    (p + "wq.bias", (q_local,), d),
                    (p + "wk.bias", (kv_local,), d),
                    (p + "wv.bias", (kv_local,), d),
                ]
            if self.qk_norm:
                specs += [
                    (p + "q_norm.weight", (hd,), d),
                    (p + "k_norm.weight", (hd,), d),
                ]
            specs += [
                (p + "wo.weight", (h, q_local), d),
                (p + "mlp_norm.weight", (h,), d),
            ]
            if self.num_experts:
                # Mixtral-family: router replicated. Default: every
                # expert Megatron-sharded (leaf names w_gate/w_up/w_down
                # keep the col/row-parallel slicing rules). EP: this
                # rank holds its share of WHOLE experts (global indices
                # preserved in the names so checkpoints map naturally).
                specs.append((p + "router.weight",
                              (self.num_experts, h), d))
                moe_i = self.moe_intermediate_size or self.intermediate_size
                for e in self.local_experts(tp_rank, tp_size):
                    ep = p + f"experts.{e}."
                    ei = moe_i if (
                        self.expert_parallel and tp_size > 1) \
                        else moe_i // tp_size
                    specs += [
                        (ep + "w_gate.weight", (ei, h), d),
                        (ep + "w_up.weight", (ei, h), d),
                        (ep + "w_down.weight", (h, ei), d),
                    ]
            else:
                specs += [
                    (p + "w_gate.weight", (i_local, h), d),
                    (p + "w_up.weight", (i_local, h), d),
                    (p + "w_down.weight", (h, i_local), d),
                ]
        specs += [
            ("final_norm.weight", (h,), d),
            ("lm_head.weight", (self.vocab_size, h), d),
        ]
        return specs

    def total_param_bytes(self, tp_rank: int = 0, tp_size: int = 1) -> int:
        esize = torch.empty(0, dtype=self.dtype).element_size()
        return sum(esize * math.prod(s) for _, s, _ in
                   self.param_specs(tp_rank, tp_size))


def _pick_next(logits: torch.Tensor, temperature: float,
               top_p: float) -> torch.Tensor:
    """[B, V] -> [B, 1] next-token ids; temperature 0 = greedy, else
    nucleus sampling."""
    if temperature <= 0.0:
        return logits.argmax(-1, keepdim=True)
    probs = torch.softmax(logits.float() / temperature, dim=-1)
    if top_p < 1.0:
        sp, si = probs.sort(dim=-1, descending=True)
        cum = sp.cumsum(-1)
        # keep the smallest prefix whose mass reaches top_p
        cut = cum - sp >= top_p
        sp = sp.masked_fill(cut, 0.0)
        sp = sp / sp.sum(-1, keepdim=True)
        pick = torch.multinomial(sp, 1)
        return si.gather(-1, pick)
    return torch.multinomial(probs, 1)


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * w.float()).to(x.dtype)


def _scale_inv_freq(inv: torch.Tensor, scaling: dict) -> torch.Tensor:
    """HF rope_scaling: "linear" divides every frequency by factor;
    "llama3" (llama-3.1+) keeps high frequencies, divides low ones and
    smoothly interpolates between (the transformers rule)."""
    rtype = scaling.get("rope_type", scaling.get("type", ""))
    factor = float(scaling.get("factor", 1.0))
    if rtype == "linear":
        return inv / factor
    if rtype == "llama3":
        low = float(scaling.get("low_freq_factor", 1.0))
        high = float(scaling.get("high_freq_factor", 4.0))
        orig = float(scaling.get("original_max_position_embeddings", 8192))
        import math as _m
        wavelen = 2 * _m.pi / inv
        low_wl = orig / low
        high_wl = orig / high
        out = torch.where(wavelen > low_wl, inv / factor, inv)
        smooth = (orig / wavelen - low) / (high - low)
        smoothed = (1 - smooth) * out / factor + smooth * out
        mid = (wavelen >= high_wl) & (wavelen <= low_wl)
        return torch.where(mid, smoothed, out)
    if rtype in ("", "default"):
        return inv
    raise ValueError(f"unsupported rope_scaling type {rtype!r}")


def precompute_rope(cfg: LlamaConfig, device) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-precomputed cos/sin tables (trig on device turns a memory-bound
    op VALU-bound — cdna_hip_programming.md Appendix B / element-wise)."""
    hd = cfg.head_dim
    inv = 1.0 / (cfg.rope_theta ** (torch.arange(0, hd, 2, dtype=torch.float32) / hd))
    if cfg.rope_scaling:
        inv = _scale_inv_freq(inv, cfg.rope_scaling)
    t = torch.arange(cfg.max_seq_len, dtype=torch.float32)
    freqs = torch.outer(t, inv)
    return freqs.cos().to(device), freqs.sin().to(device)


def apply_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               start_pos: int) -> torch.Tensor:
    # x: [B, T, heads, head_dim]
    B, T, H, D = x.shape
    c = cos[start_pos:start_pos + T].view(1, T, 1, D // 2)
    s = sin[start_pos:start_pos + T].view(1, T, 1, D // 2)
    xf = x.float().view(B, T, H, D // 2, 2)
    x0, x1 = xf[..., 0], xf[..., 1]
    out = torch.stack((x0 * c - x1 * s, x0 * s + x1 * c), dim=-1)
    return out.view(B, T, H, D).to(x.dtype)


class KVCache:
    def __init__(self, cfg: LlamaConfig, batch: int, device,
                 tp_size: int = 1, max_seq: Optional[int] = None):
        kv_heads = max(cfg.num_kv_heads // tp_size, 1)
        seq = max_seq or cfg.max_seq_len
        shape = (cfg.num_layers, 2, batch, seq, kv_heads, cfg.head_dim)
        self.data = torch.zeros(shape, dtype=cfg.dtype, device=device)
        self.seq_len = 0

    def free(self) -> None:
        self.data = None  # type: ignore[assignment]


class LlamaModel:
    """Functional Llama over a dict of (arena-view) tensors."""

    def __init__(self, cfg: LlamaConfig, params: Dict[str, torch.Tensor],
                 device, tp_rank: int = 0, tp_size: int = 1,
                 tp_group: Optional[object] = None):
        self.cfg = cfg
        self.params = params
        self.device = device
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self.tp_group = tp_group
        self.rope_cos, self.rope_sin = precompute_rope(cfg, device)

    def rebind(self, params: Dict[str, torch.Tensor]) -> None:
        """Swap in fresh views after a non-VMM wake changed the arena base."""
        self.params = params

    def init_weights(self, seed: int = 0) -> None:
        import zlib
        torch.manual_seed(seed + self.tp_rank)
        kv_rep = max(self.tp_size // self.cfg.num_kv_heads, 1)
        for name, p in self.params.items():
            if p.dim() > 1:
                if kv_rep > 1 and (".wk." in name or ".wv." in name):
                    # KV replication (tp > kv heads): ranks sharing a
                    # logical kv head must generate identical weights
                    group = self.tp_rank * self.cfg.num_kv_heads \
                        // self.tp_size
                    g = torch.Generator(device=p.device)
                    g.manual_seed((seed * 1000003 + group) * 131071
                                  + zlib.crc32(name.encode()))
                    p.normal_(0.0, 0.02, generator=g)
                else:
                    p.normal_(0.0, 0.02)  # in-place device RNG: no temps
            elif name.endswith(".bias"):
                p.normal_(0.0, 0.02)
            else:
                p.fill_(1.0)  # norm gains

    def _act(self, gate: torch.Tensor, up: torch.Tensor,
             decode1: bool) -> torch.Tensor:
        """Gated-MLP activation: fused silu kernel on the decode path,
        gelu-tanh (gemma) via torch — the GEMVs dominate either way."""
        if self.cfg.hidden_act == "silu":
            return fast_silu_mul(gate, up) if decode1                 else F.silu(gate) * up
        if self.cfg.hidden_act == "gelu_tanh":
            return F.gelu(gate, approximate="tanh") * up
        raise ValueError(f"unknown hidden_act {self.cfg.hidden_act!r}")

    def _maybe_all_reduce(self, x: torch.Tensor) -> torch.Tensor:
        if self.tp_size > 1:
            dist.all_reduce(x, group=self.tp_group)
        return x

    def _moe_mlp(self, x: torch.Tensor, P, p: str,
                 decode1: bool) -> torch.Tensor:
        """Mixtral-style sparse MLP: softmax(top-k(router)) over
        Megatron-sharded experts; ONE all-reduce after the weighted sum
        (row-parallel partials add across ranks exactly like the dense
        w_down). Router weights are replicated so every rank routes
        identically. The decode path runs only the k selected experts'
        GEMVs — at batch 1 a sparse model moves k/E of the expert bytes."""
        cfg = self.cfg
        h = fast_rmsnorm(x, P[p + "mlp_norm.weight"], cfg.norm_eps) \
            if decode1 else rmsnorm(x, P[p + "mlp_norm.weight"],
                                    cfg.norm_eps)
        B, T, H = h.shape
        logits = fast_linear(h, P[p + "router.weight"]).float()
        if cfg.moe_norm_topk:
            topw, topi = torch.topk(logits, cfg.num_experts_per_tok,
                                    dim=-1)
            topw = torch.softmax(topw, dim=-1)
        else:
            # qwen3-moe norm_topk_prob=false: global softmax, top-k
            # probabilities used as-is (they do not sum to 1)
            probs = torch.softmax(logits, dim=-1)
            topw, topi = torch.topk(probs, cfg.num_experts_per_tok,
                                    dim=-1)
        mine = set(cfg.local_experts(self.tp_rank, self.tp_size))
        if B * T == 1:
            out = torch.zeros_like(x)
            idx = topi.reshape(-1).tolist()  # k tiny ints; eager decode
            for j, e in enumerate(idx):
                if e not in mine:
                    continue  # EP: another rank owns it; all-reduce sums
                ep = p + f"experts.{e}."
                gate, up = fast_linear_multi(
                    h, (P[ep + "w_gate.weight"], P[ep + "w_up.weight"]))
                act = self._act(gate, up, decode1)
                w = topw.reshape(-1)[j].to(x.dtype)
                out = out + w * fast_linear(act, P[ep + "w_down.weight"])
        else:
            hf = h.view(B * T, H)
            ti = topi.view(B * T, -1)
            tw = topw.view(B * T, -1)
            out = torch.zeros(B * T, H, dtype=torch.float32,
                              device=x.device)
            for e in mine:
                mask = ti == e
                tok = mask.any(-1).nonzero(as_tuple=True)[0]
                if tok.numel() == 0:
                    continue
                ep = p + f"experts.{e}."
                sub = hf[tok]
                act = F.silu(F.linear(sub, P[ep + "w_gate.weight"])) * \
                    F.linear(sub, P[ep + "w_up.weight"])
                y = F.linear(act, P[ep + "w_down.weight"]).float()
                w = (tw * mask).sum(-1)[tok].unsqueeze(-1)
                out.index_add_(0, tok, w * y)
            out = out.to(x.dtype).view(B, T, H)
        return x + self._maybe_all_reduce(out)

    @torch.no_grad()
    def forward(self, tokens: torch.Tensor, cache: Optional[KVCache] = None,
                start_pos: int = 0) -> torch.Tensor:
        """tokens [B, T] -> logits [B, T, vocab]."""
        cfg = self.cfg
        P = self.params
        B, T = tokens.shape
        q_heads = cfg.num_heads // self.tp_size
        kv_heads = max(cfg.num_kv_heads // self.tp_size, 1)
        hd = cfg.head_dim

        x = F.embedding(tokens, P["embed.weight"])
        if cfg.embed_scale:  # gemma: normalizer on the residual stream
            x = x * (cfg.hidden_size ** 0.5)
        # fused single-token decode path (one kernel per elementwise op)
        decode1 = (B == 1 and T == 1 and x.is_cuda
                   and x.dtype == torch.bfloat16)
        for li in range(cfg.num_layers):
            p = f"layers.{li}."
            biases = (P[p + "wq.bias"], P[p + "wk.bias"],
                      P[p + "wv.bias"]) if cfg.qkv_bias else None
            if decode1:
                # qkv in ONE launch, biases fused into the GEMV stores.
                # (A norm-fused variant exists — gemv_multi_bf16(
                # norm_w=...) — but measured SLOWER: every block
                # redundantly re-reads x and reduces the sum of squares,
                # costing more than the one saved launch: 228 -> 220
                # tok/s short ctx. Selection is measured.)
                h = fast_rmsnorm(x, P[p + "attn_norm.weight"],
                                 cfg.norm_eps)
                q, k, v = fast_linear_multi(
                    h, (P[p + "wq.weight"], P[p + "wk.weight"],
                        P[p + "wv.weight"]), biases=biases)
            else:
                h = rmsnorm(x, P[p + "attn_norm.weight"], cfg.norm_eps)
                q = fast_linear(h, P[p + "wq.weight"])
                k = fast_linear(h, P[p + "wk.weight"])
                v = fast_linear(h, P[p + "wv.weight"])
                if biases is not None:
                    q = q + biases[0]
                    k = k + biases[1]
                    v = v + biases[2]
            q = q.view(B, T, q_heads, hd)
            k = k.view(B, T, kv_heads, hd)
            v = v.view(B, T, kv_heads, hd)
            if cfg.qk_norm:
                q = rmsnorm(q, P[p + "q_norm.weight"], cfg.norm_eps)
                k = rmsnorm(k, P[p + "k_norm.weight"], cfg.norm_eps)
            stored = False
            if decode1 and cache is not None and not cfg.qk_norm:
                # RoPE(q) + RoPE(k)->cache + v->cache in ONE launch
                stored = fast_rope_qkv_store(
                    q, k, v, cache.data[li, 0, 0], cache.data[li, 1, 0],
                    self.rope_cos, self.rope_sin, pos=start_pos)
            if not stored:
                if decode1:
                    q = fast_rope1(q, self.rope_cos[start_pos],
                                   self.rope_sin[start_pos])
                    k = fast_rope1(k, self.rope_cos[start_pos],
                                   self.rope_sin[start_pos])
                else:
                    q = apply_rope(q, self.rope_cos, self.rope_sin,
                                   start_pos)
                    k = apply_rope(k, self.rope_cos, self.rope_sin,
                                   start_pos)
            if cache is not None:
                if not stored:
                    cache.data[li, 0, :, start_pos:start_pos + T] = k
                    cache.data[li, 1, :, start_pos:start_pos + T] = v
                k = cache.data[li, 0, :, : start_pos + T]
                v = cache.data[li, 1, :, : start_pos + T]
            att = None
            if decode1 and cache is not None:
                # fused GQA decode attention over the raw cache (no
                # repeat_interleave copies, one launch)
                att = fast_attn_decode(q, cache.data[li, 0, 0],
                                       cache.data[li, 1, 0], start_pos + 1)
            if att is None and B == 1 and T > 1 and x.is_cuda \
                    and x.dtype == torch.bfloat16:
                # causal GQA prefill on MFMA matrix cores (one launch,
                # reads the cache layout directly)
                if cache is not None:
                    att = fast_attn_prefill(q, cache.data[li, 0, 0],
                                            cache.data[li, 1, 0], start_pos)
                elif start_pos == 0:
                    att = fast_attn_prefill(q, k[0], v[0], 0)
            if att is None:
                # SDPA wants [B, heads, T, hd]
                qh = q.transpose(1, 2)
                kh = k.transpose(1, 2)
                vh = v.transpose(1, 2)
                if kv_heads != q_heads:
                    rep = q_heads // kv_heads
                    kh = kh.repeat_interleave(rep, dim=1)
                    vh = vh.repeat_interleave(rep, dim=1)
                att = F.scaled_dot_product_attention(
                    qh, kh, vh, is_causal=(T > 1))
                att = att.transpose(1, 2).reshape(B, T, q_heads * hd)
            if decode1 and self.tp_size == 1:
                x = fast_linear_residual(att, P[p + "wo.weight"], x)
            else:
                x = x + self._maybe_all_reduce(
                    fast_linear(att, P[p + "wo.weight"]))

            if cfg.num_experts:
                x = self._moe_mlp(x, P, p, decode1)
                continue
            if decode1:
                h = fast_rmsnorm(x, P[p + "mlp_norm.weight"],
                                 cfg.norm_eps)
                gate, up = fast_linear_multi(
                    h, (P[p + "w_gate.weight"], P[p + "w_up.weight"]))
            else:
                h = rmsnorm(x, P[p + "mlp_norm.weight"], cfg.norm_eps)
                gate = fast_linear(h, P[p + "w_gate.weight"])
                up = fast_linear(h, P[p + "w_up.weight"])
            # NOTE: a silu-fused w_down GEMV exists (gemv_silu_bf16 /
            # fast_down_proj) but measured SLOWER: every GEMV block
            # recomputes silu over its x stage (239.6 -> 228 tok/s), and
            # on wide-K models whose x exceeds the LDS stage it
            # recomputes per dot element (70B-shape 74.7 -> 60.5).
            # Selection is measured; the separate activation launch wins.
            act = self._act(gate, up, decode1)
            if decode1 and self.tp_size == 1:
                x = fast_linear_residual(act, P[p + "w_down.weight"], x)
            else:
                x = x + self._maybe_all_reduce(
                    fast_linear(act, P[p + "w_down.weight"]))

        x = fast_rmsnorm(x, P["final_norm.weight"], cfg.norm_eps) \
            if decode1 else rmsnorm(x, P["final_norm.weight"], cfg.norm_eps)
        return fast_linear(x, P["lm_head.weight"]).float()

    @torch.no_grad()
    def generate(self, prompt: torch.Tensor, max_new_tokens: int = 16,
                 cache: Optional[KVCache] = None,
                 eos_id: Optional[int] = None,
                 temperature: float = 0.0,
                 top_p: float = 1.0) -> torch.Tensor:
        """Decode. prompt [B, T] -> [B, T + up to max_new_tokens];
        greedy at temperature 0, else nucleus sampling (top_p) from
        softmax(logits / temperature). Stops early once every sequence
        emitted ``eos_id``."""
        B, T = prompt.shape
        own_cache = cache is None
        if own_cache:
            cache = KVCache(self.cfg, B, self.device, self.tp_size,
                            max_seq=min(self.cfg.max_seq_len,
                                        T + max_new_tokens))
        logits = self.forward(prompt, cache, 0)
        out = prompt
        pos = T
        done = torch.zeros(B, dtype=torch.bool, device=prompt.device)
        for _ in range(max_new_tokens):
            nxt = _pick_next(logits[:, -1], temperature, top_p)
            out = torch.cat([out, nxt], dim=1)
            if eos_id is not None:
                done |= nxt[:, 0] == eos_id
                if bool(done.all()):
                    break
            logits = self.forward(nxt, cache, pos)
            pos += 1
        if own_cache:
            cache.free()
        return out
