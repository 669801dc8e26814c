"""MFMA prefill attention kernel vs torch SDPA: latency at llama shapes.

Usage (GPU box):  python tools/prefill_bench.py [--iters 50]
Measures the attention op alone (8B shape: qH=32, kvH=8, hd=128) and the
full-model prefill (tiny model) with FMA_DISABLE_MFMA_PREFILL toggled.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def bench_op(fn, iters):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--t", type=int, default=0,
                    help="single T, MFMA kernel only (for profiling)")
    args = ap.parse_args()
    import fma_amd._C as C

    qH, kvH, hd = 32, 8, 128
    rep = qH // kvH
    if args.t:
        T = args.t
        q = torch.randn(T, qH, hd, dtype=torch.bfloat16, device="cuda:0")
        k = torch.randn(T, kvH, hd, dtype=torch.bfloat16, device="cuda:0")
        v = torch.randn(T, kvH, hd, dtype=torch.bfloat16, device="cuda:0")
        t_ms = bench_op(lambda: C.attn_prefill_bf16(q, k, v, 0), args.iters)
        print(f"T={T} mfma_ms={t_ms:.3f}")
        return
    print(f"{'T':>6} {'mfma_ms':>9} {'m16_ms':>9} {'sdpa_ms':>9} "
          f"{'sdpa/m16':>8} {'max_err16':>9}")
    for T in (128, 256, 512, 1024, 2048, 4096):
        q = torch.randn(T, qH, hd, dtype=torch.bfloat16, device="cuda:0")
        k = torch.randn(T, kvH, hd, dtype=torch.bfloat16, device="cuda:0")
        v = torch.randn(T, kvH, hd, dtype=torch.bfloat16, device="cuda:0")
        qh = q.permute(1, 0, 2).unsqueeze(0).contiguous()
        kh = k.permute(1, 0, 2).repeat_interleave(rep, 0).unsqueeze(0).contiguous()
        vh = v.permute(1, 0, 2).repeat_interleave(rep, 0).unsqueeze(0).contiguous()

        t_mfma = bench_op(lambda: C.attn_prefill_bf16(q, k, v, 0), args.iters)
        # 16-row variant (chunks=1 forced so the env gate applies)
        os.environ["FMA_PREFILL_16"] = "1"
        t_m16 = bench_op(lambda: C.attn_prefill_bf16(q, k, v, 0, 1),
                         args.iters)
        out16 = C.attn_prefill_bf16(q, k, v, 0, 1).float()
        os.environ.pop("FMA_PREFILL_16", None)
        t_sdpa = bench_op(
            lambda: torch.nn.functional.scaled_dot_product_attention(
                qh, kh, vh, is_causal=True), args.iters)

        ref = torch.nn.functional.scaled_dot_product_attention(
            qh.float(), kh.float(), vh.float(), is_causal=True)
        err16 = (out16 - ref.squeeze(0).permute(1, 0, 2)).abs().max().item()
        print(f"{T:>6} {t_mfma:>9.3f} {t_m16:>9.3f} {t_sdpa:>9.3f} "
              f"{t_sdpa / t_m16:>8.2f} {err16:>9.4f}")


if __name__ == "__main__":
    main()
