"""Fused decode elementwise ops with eager fallbacks.

One launch each for rmsnorm / silu*up / RoPE on the single-token decode
path (profiling showed the fragmented eager versions cost more GPU time
than the GEMVs). Fallbacks implement identical math for CPU and for
non-decode shapes; numerics are compared in tests/gpu.
"""

from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from fma_amd.ops import actuation

_ENABLED = None


def available() -> bool:
    global _ENABLED
    if _ENABLED is None:
        _ENABLED = (torch.cuda.is_available()
                    and actuation.native_available()
                    and os.environ.get("FMA_DISABLE_FUSED_OPS") != "1")
    return _ENABLED


def _fusable(x: torch.Tensor) -> bool:
    return (available() and x.is_cuda and x.dtype == torch.bfloat16
            and x.is_contiguous() and (x.numel() & 7) == 0)


def fast_rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    """Single-token rmsnorm; x may be [..., H] with numel == H."""
    H = x.shape[-1]
    if _fusable(x) and x.numel() == H and w.is_contiguous():
        return actuation._C.rmsnorm1_bf16(x.reshape(-1), w, eps).view(x.shape)
    xf = x.float()
    xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * w.float()).to(x.dtype)


def fast_silu_mul(g: torch.Tensor, u: torch.Tensor) -> torch.Tensor:
    if _fusable(g) and u.is_contiguous():
        return actuation._C.silu_mul_bf16(
            g.reshape(-1), u.reshape(-1)).view(g.shape)
    return F.silu(g) * u


def fast_rope1(x: torch.Tensor, cos_row: torch.Tensor, sin_row: torch.Tensor
               ) -> torch.Tensor:
    """RoPE for one position: x [B=1, 1, heads, hd]; cos/sin_row [hd/2] fp32.
    In-place on the fused path (x is a fresh projection output)."""
    B, T, heads, hd = x.shape
    if _fusable(x) and B == 1 and T == 1:
        actuation._C.rope1_bf16_(x.reshape(-1), cos_row.reshape(-1),
                                 sin_row.reshape(-1), heads, hd)
        return x
    c = cos_row.view(1, 1, 1, hd // 2)
    s = sin_row.view(1, 1, 1, hd // 2)
    xf = x.float().view(B, T, heads, hd // 2, 2)
    x0, x1 = xf[..., 0], xf[..., 1]
    out = torch.stack((x0 * c - x1 * s, x0 * s + x1 * c), dim=-1)
    return out.view(B, T, heads, hd).to(x.dtype)


def fast_rope_qkv_store(q, k, v, kcache, vcache, cos_tab, sin_tab,
                        pos=None, pos_dev=None) -> bool:
    """RoPE(q in place) + RoPE(k)->cache + v->cache in ONE launch
    (replaces 2 rope + 2 cache-write kernels on the decode path).
    q [1,1,qH,hd]; k/v [1,1,kvH,hd]; k/vcache [S,kvH,hd]. Returns False
    when the fused kernel does not apply (caller uses the unfused ops)."""
    if not available():
        return False
    if not (q.is_cuda and q.dtype == torch.bfloat16 and q.is_contiguous()
            and k.is_contiguous() and v.is_contiguous()
            and kcache.is_contiguous() and vcache.is_contiguous()):
        return False
    qh, hd = q.shape[-2], q.shape[-1]
    kvh = k.shape[-2]
    actuation._C.rope_qkv_store_bf16_(
        q.reshape(qh, hd), k.reshape(kvh, hd), v.reshape(kvh, hd),
        kcache, vcache, cos_tab, sin_tab, pos_dev,
        0 if pos is None else int(pos))
    return True


def fast_attn_prefill(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, pos0: int):
    """Causal GQA prefill on MFMA matrix cores. q [1,T,qH,hd]; k/v_cache
    [S,kvH,hd] holding keys [0, pos0+T). Returns [1,T,qH*hd] or None when
    the kernel does not apply (caller falls back to SDPA).

    Dispatch is measured, not aspirational (tools/prefill_bench.py on
    MI355X): the hand-written kernel beats torch SDPA up to T~128 and
    trails it beyond (0.56x at 4K ctx), so by default it serves short
    prefills only. FMA_MFMA_PREFILL=1 forces it always, =0 never."""
    mode = os.environ.get("FMA_MFMA_PREFILL", "auto")
    if not available() or mode == "0":
        return None
    b, t, qh, hd = q.shape
    # FMA_PREFILL_16=1 opts into the 16-row fragment kernel (occupancy 4,
    # doubled tile grid — built for exactly the mid-T range the 32-row
    # kernel loses; numerics hardware-validated, perf unmeasured), so the
    # short-prefill cutoff does not apply to it
    p16 = os.environ.get("FMA_PREFILL_16") == "1"
    if mode != "1" and not p16 and t > 128:
        return None
    if (b != 1 or hd not in (64, 128) or not q.is_cuda
            or q.dtype != torch.bfloat16
            or qh % k_cache.shape[1] != 0
            or pos0 + t > k_cache.shape[0]):
        return None
    out = actuation._C.attn_prefill_bf16(
        q.reshape(t, qh, hd).contiguous(), k_cache, v_cache, pos0)
    return out.view(1, t, qh * hd)


def fast_attn_decode(q: torch.Tensor, k_cache: torch.Tensor,
                     v_cache: torch.Tensor, t: int):
    """q [1,1,qH,hd]; k/v_cache [S,kvH,hd] (one layer, batch 1). Returns
    [1,1,qH*hd] or None when the fused kernel does not apply."""
    if not available():
        return None
    _, _, qh, hd = q.shape
    if (hd % 64 != 0 or hd > 256 or t > (1 << 17) or not q.is_cuda
            or q.dtype != torch.bfloat16
            or qh % k_cache.shape[1] != 0):
        return None
    out = actuation._C.attn_decode_bf16(
        q.reshape(qh, hd).contiguous(), k_cache, v_cache, t)
    return out.view(1, 1, qh * hd)
