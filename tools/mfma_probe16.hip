// Standalone MFMA fragment-layout probe for v_mfma_f32_16x16x32_bf16 —
// groundwork for the 16x16-fragment prefill rewrite (NOTES.md round-3
// design sketch: halves oacc/qf VGPRs vs 32x32x16, doubling occupancy
// headroom at mid-T). Computes D = A(16x32) * B(32x16) with the assumed
// lane->element mapping and prints mismatches vs a CPU reference.
// Build & run on a GPU box:
//   hipcc --offload-arch=gfx950 -O2 tools/mfma_probe16.hip -o /tmp/p16 && /tmp/p16
//
// Assumed mapping (guide cdna_hip_programming.md §3: C/D is
// col=lane&15, row=(lane>>4)*4+reg; A/B by natural extension of the
// 32x32x16 input layout — 4 lane-groups of 16 cover K=32 in 8-element
// runs):
//   A: lane l holds A[l%16][8*(l/16) + i], i in [0,8)  (8 bf16 = 4 VGPRs)
//   B: lane l holds B[8*(l/16) + i][l%16]
//   C/D: lane l holds D[4*(l/16) + reg][l%16], reg in [0,4)

#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__global__ void probe(const __bf16* A, const __bf16* B, float* D) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  for (int i = 0; i < 8; ++i) {
    a[i] = A[(l % 16) * 32 + (8 * (l / 16) + i)];
    b[i] = B[(8 * (l / 16) + i) * 16 + (l % 16)];
  }
  f32x4 c{};
  for (int i = 0; i < 4; ++i) c[i] = 0.0f;
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  for (int reg = 0; reg < 4; ++reg) {
    const int row = 4 * (l / 16) + reg;
    const int col = l % 16;
    D[row * 16 + col] = c[reg];
  }
}

int main() {
  std::vector<float> Ah(16 * 32), Bh(32 * 16);
  for (int i = 0; i < 16; ++i)
    for (int k = 0; k < 32; ++k) Ah[i * 32 + k] = 0.01f * i + 0.25f * k;
  for (int k = 0; k < 32; ++k)
    for (int j = 0; j < 16; ++j) Bh[k * 16 + j] = 0.02f * j - 0.15f * k;
  std::vector<__bf16> Ab(16 * 32), Bb(32 * 16);
  for (int i = 0; i < 16 * 32; ++i) Ab[i] = static_cast<__bf16>(Ah[i]);
  for (int i = 0; i < 32 * 16; ++i) Bb[i] = static_cast<__bf16>(Bh[i]);

  __bf16 *dA, *dB;
  float* dD;
  hipMalloc(&dA, sizeof(__bf16) * 16 * 32);
  hipMalloc(&dB, sizeof(__bf16) * 32 * 16);
  hipMalloc(&dD, sizeof(float) * 16 * 16);
  hipMemcpy(dA, Ab.data(), sizeof(__bf16) * 16 * 32, hipMemcpyHostToDevice);
  hipMemcpy(dB, Bb.data(), sizeof(__bf16) * 32 * 16, hipMemcpyHostToDevice);
  probe<<<1, 64>>>(dA, dB, dD);
  hipDeviceSynchronize();
  std::vector<float> Dh(16 * 16);
  hipMemcpy(Dh.data(), dD, sizeof(float) * 16 * 16, hipMemcpyDeviceToHost);

  int bad = 0;
  float max_err = 0.0f;
  for (int i = 0; i < 16; ++i) {
    for (int j = 0; j < 16; ++j) {
      float ref = 0.0f;
      for (int k = 0; k < 32; ++k) {
        // bf16-rounded inputs for a fair reference
        ref += static_cast<float>(static_cast<__bf16>(Ah[i * 32 + k])) *
               static_cast<float>(static_cast<__bf16>(Bh[k * 16 + j]));
      }
      const float got = Dh[i * 16 + j];
      const float err = fabsf(got - ref);
      max_err = fmaxf(max_err, err);
      if (err > 0.05f && bad < 8) {
        printf("MISMATCH D[%d][%d]: got %f ref %f\n", i, j, got, ref);
        ++bad;
      }
    }
  }
  printf(bad ? "LAYOUT WRONG (max_err=%f)\n" : "LAYOUT OK (max_err=%f)\n",
         max_err);
  return bad ? 1 : 0;
}
