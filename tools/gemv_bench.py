"""GEMV kernel vs hipBLASLt (F.linear) on llama decode shapes."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

import fma_amd._C as C  # noqa: E402


def bench(M, K, reps=50):
    W = torch.randn(M, K, dtype=torch.bfloat16, device="cuda:0")
    x = torch.randn(K, dtype=torch.bfloat16, device="cuda:0")
    # numerics vs fp32 reference
    ref = (W.float() @ x.float())
    out = C.gemv_bf16(W, x)
    err = (out - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
    torch.cuda.synchronize()
    import time
    for fn, name in ((lambda: C.gemv_bf16(W, x), "fma_gemv"),
                     (lambda: F.linear(x, W), "hipblaslt")):
        fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / reps
        bw = M * K * 2 / dt / 1e12
        print(f"M={M:6d} K={K:5d} {name:10s} {dt*1e6:8.1f} us  {bw:5.2f} TB/s"
              + (f"  rel_err={err:.2e}" if name == "fma_gemv" else ""))


if __name__ == "__main__":
    for M, K in ((4096, 4096), (14336, 4096), (4096, 14336),
                 (28672, 8192), (8192, 28672), (32768, 4096),
                 (128256, 8192)):
        bench(M, K)
