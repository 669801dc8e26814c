// Standalone correctness check for the 16-row MFMA prefill kernel
// (attn_prefill16_bf16_kernel in fma_amd/csrc/kernels.hip) — links the
// EXACT integrated kernel and compares against a CPU reference on
// causal+GQA shapes covering both NW paths, ragged T, pos0>0, both HDs.
// Build & run on a GPU box:
//   hipcc --offload-arch=gfx950 -O2 tools/prefill16_probe.hip \
//       fma_amd/csrc/kernels.hip -o /tmp/pp16 && /tmp/pp16
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <vector>

extern "C" hipError_t fma_launch_attn_prefill16_bf16(
    const void* Q, const void* K, const void* V, void* O, int T, int pos0,
    int q_heads, int kv_heads, int hd, hipStream_t stream);

static unsigned lcg_state = 12345u;
static float frand() {  // deterministic, in [-1, 1)
  lcg_state = lcg_state * 1664525u + 1013904223u;
  return static_cast<float>(static_cast<int>(lcg_state >> 9) & 0xFFFF) /
             32768.0f -
         1.0f;
}

static unsigned short f2b(float f) {
  union {
    float f;
    unsigned u;
  } v{f};
  const unsigned r = v.u + 0x7FFF + ((v.u >> 16) & 1);
  return static_cast<unsigned short>(r >> 16);
}
static float b2f(unsigned short h) {
  union {
    unsigned u;
    float f;
  } v;
  v.u = static_cast<unsigned>(h) << 16;
  return v.f;
}

// one (T, pos0, qH, kvH, hd) case; returns max abs error
static float run_case(int T, int pos0, int qH, int kvH, int hd) {
  const int S = pos0 + T;
  std::vector<unsigned short> Q(static_cast<size_t>(T) * qH * hd),
      K(static_cast<size_t>(S) * kvH * hd), V(K.size()), O(Q.size(), 0);
  for (auto& x : Q) x = f2b(frand());
  for (auto& x : K) x = f2b(frand());
  for (auto& x : V) x = f2b(frand());

  unsigned short *dQ, *dK, *dV, *dO;
  hipMalloc(&dQ, Q.size() * 2);
  hipMalloc(&dK, K.size() * 2);
  hipMalloc(&dV, V.size() * 2);
  hipMalloc(&dO, O.size() * 2);
  hipMemcpy(dQ, Q.data(), Q.size() * 2, hipMemcpyHostToDevice);
  hipMemcpy(dK, K.data(), K.size() * 2, hipMemcpyHostToDevice);
  hipMemcpy(dV, V.data(), V.size() * 2, hipMemcpyHostToDevice);
  hipError_t err =
      fma_launch_attn_prefill16_bf16(dQ, dK, dV, dO, T, pos0, qH, kvH, hd,
                                     nullptr);
  hipDeviceSynchronize();
  if (err != hipSuccess) {
    printf("LAUNCH FAILED: %s\n", hipGetErrorString(err));
    return 1e9f;
  }
  hipMemcpy(O.data(), dO, O.size() * 2, hipMemcpyDeviceToHost);
  hipFree(dQ);
  hipFree(dK);
  hipFree(dV);
  hipFree(dO);

  const float scale = 1.0f / sqrtf(static_cast<float>(hd));
  const int group = qH / kvH;
  float max_err = 0.0f;
  for (int t = 0; t < T; ++t) {
    const int qpos = pos0 + t;
    for (int h = 0; h < qH; ++h) {
      const int kh = h / group;
      // scores + online-free softmax in double
      std::vector<double> sc(qpos + 1);
      double m = -1e30;
      for (int k = 0; k <= qpos; ++k) {
        double s = 0.0;
        for (int d = 0; d < hd; ++d) {
          s += static_cast<double>(
                   b2f(Q[(static_cast<size_t>(t) * qH + h) * hd + d])) *
               b2f(K[(static_cast<size_t>(k) * kvH + kh) * hd + d]);
        }
        sc[k] = s * scale;
        if (sc[k] > m) m = sc[k];
      }
      double l = 0.0;
      for (int k = 0; k <= qpos; ++k) {
        sc[k] = exp(sc[k] - m);
        l += sc[k];
      }
      for (int d = 0; d < hd; ++d) {
        double o = 0.0;
        for (int k = 0; k <= qpos; ++k) {
          o += sc[k] *
               b2f(V[(static_cast<size_t>(k) * kvH + kh) * hd + d]);
        }
        const float ref = static_cast<float>(o / l);
        const float got =
            b2f(O[(static_cast<size_t>(t) * qH + h) * hd + d]);
        const float err2 = fabsf(got - ref);
        if (err2 > max_err) max_err = err2;
      }
    }
  }
  return max_err;
}

int main() {
  struct Case {
    int T, pos0, qH, kvH, hd;
    const char* label;
  } cases[] = {
      {64, 0, 8, 2, 128, "hd128 NW4 aligned"},
      {100, 28, 8, 2, 128, "hd128 NW4 ragged+pos0"},
      {50, 0, 6, 2, 128, "hd128 NW1 (group 3)"},
      {80, 16, 8, 2, 64, "hd64 NW4"},
      {33, 5, 3, 3, 64, "hd64 NW1 MHA ragged"},
  };
  int bad = 0;
  for (const auto& cs : cases) {
    const float e = run_case(cs.T, cs.pos0, cs.qH, cs.kvH, cs.hd);
    const bool ok = e < 0.035f;  // bf16 P + bf16 inputs tolerance
    printf("%-24s max_err=%f %s\n", cs.label, e, ok ? "OK" : "FAIL");
    if (!ok) ++bad;
  }
  printf(bad ? "PREFILL16 WRONG\n" : "PREFILL16 OK\n");
  return bad ? 1 : 0;
}
