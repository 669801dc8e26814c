"""Linear dispatch: hand-written GEMV for batch-1 decode, hipBLASLt else.

Batch-1 decode reads every weight byte once per token — pure streaming.
hipBLASLt's batch-1 GEMV measured ~1 TB/s effective on llama shapes;
csrc/kernels.hip's wave-per-row GEMV targets the streaming ceiling. The
dispatch is shape-based: single-token inputs on bf16 weights take the
kernel, everything else (prefill, batches) stays on hipBLASLt GEMMs.
"""

from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from fma_amd.ops import actuation

_ENABLED = None


def gemv_enabled() -> bool:
    global _ENABLED
    if _ENABLED is None:
        _ENABLED = (torch.cuda.is_available()
                    and actuation.native_available()
                    and os.environ.get("FMA_DISABLE_GEMV") != "1")
    return _ENABLED


def fast_linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """F.linear(x, w) with the decode-GEMV fast path."""
    if (gemv_enabled() and x.is_cuda
            and w.dtype == torch.bfloat16 and x.dtype == torch.bfloat16
            and x.numel() == x.shape[-1]          # B*T == 1
            and (w.shape[1] & 7) == 0
            and w.is_contiguous()):
        y = actuation._C.gemv_bf16(w, x.reshape(-1).contiguous(),
                                   True)  # bf16 out, convert fused
        return y.view(*x.shape[:-1], w.shape[0])
    return F.linear(x, w)


def fast_linear_multi(x, weights, norm=None, biases=None):
    """1-3 projections of the same single-token x in ONE kernel launch
    (qkv, gate+up), optionally with per-weight biases fused into the
    stores (Qwen2-family qkv biases) and/or the rmsnorm producing the
    GEMV input fused in (norm = (weight, eps); x is then the RAW
    residual stream). Falls back to eager math off the fast path."""
    K = x.shape[-1]
    if (gemv_enabled() and x.is_cuda
            and x.dtype == torch.bfloat16
            and x.numel() == K
            and (norm is None or (norm[0].is_contiguous()
                                  and norm[0].dtype == torch.bfloat16
                                  and K * 2 <= 32 * 1024))
            and (biases is None
                 or all(b.dtype == torch.bfloat16 and b.is_contiguous()
                        for b in biases))
            and all(w.dtype == torch.bfloat16 and w.is_contiguous()
                    and (w.shape[1] & 7) == 0 for w in weights)):
        blist = [b.reshape(-1) for b in biases] if biases is not None             else None
        if norm is None:
            ys = actuation._C.gemv_multi_bf16(
                x.reshape(-1).contiguous(), list(weights), biases=blist)
        else:
            ys = actuation._C.gemv_multi_bf16(
                x.reshape(-1).contiguous(), list(weights),
                norm[0].reshape(-1), float(norm[1]), biases=blist)
        return [y.view(*x.shape[:-1], w.shape[0])
                for y, w in zip(ys, weights)]
    if norm is not None:
        xf = x.float()
        xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + norm[1])
        x = (xf * norm[0].float()).to(x.dtype)
    outs = [F.linear(x, w) for w in weights]
    if biases is not None:
        outs = [o + b for o, b in zip(outs, biases)]
    return outs


def fast_down_proj(gate, up, w, residual):
    """residual + F.linear(silu(gate)*up, w) with the activation fused
    into the GEMV input stage (one launch for the whole MLP tail)."""
    if (gemv_enabled() and gate.is_cuda
            and gate.dtype == torch.bfloat16
            and gate.numel() == gate.shape[-1]
            and up.is_contiguous()
            and residual.numel() == w.shape[0]
            and (w.shape[1] & 7) == 0 and w.is_contiguous()):
        y = actuation._C.gemv_silu_bf16(
            w, gate.reshape(-1).contiguous(), up.reshape(-1).contiguous(),
            residual.reshape(-1).contiguous())
        return y.view(residual.shape)
    act = F.silu(gate) * up
    return residual + F.linear(act, w)


def fast_linear_residual(x: torch.Tensor, w: torch.Tensor,
                         residual: torch.Tensor) -> torch.Tensor:
    """residual + F.linear(x, w) with the add fused into the GEMV store."""
    if (gemv_enabled() and x.is_cuda
            and w.dtype == torch.bfloat16 and x.dtype == torch.bfloat16
            and x.numel() == x.shape[-1]
            and residual.numel() == w.shape[0]
            and (w.shape[1] & 7) == 0
            and w.is_contiguous()):
        y = actuation._C.gemv_bf16(w, x.reshape(-1).contiguous(), True,
                                   residual.reshape(-1).contiguous())
        return y.view(residual.shape)
    return residual + F.linear(x, w)
