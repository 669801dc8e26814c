// Standalone MFMA fragment-layout probe for v_mfma_f32_32x32x16_bf16.
// Computes D = A(32x16) * B(16x32) with an assumed lane->element mapping
// and prints mismatches vs a CPU reference. Build & run on a GPU box:
//   hipcc --offload-arch=gfx950 -O2 tools/mfma_probe.hip -o /tmp/probe && /tmp/probe
//
// Assumed mapping (natural extension of the CDNA3 32x32x8 layout):
//   A: lane l holds A[l%32][8*(l/32) + i], i in [0,8)   (8 bf16 = 4 VGPRs)
//   B: lane l holds B[8*(l/32) + i][l%32]
//   C/D: lane l holds D[(reg%4) + 8*(reg/4) + 4*(l/32)][l%32], reg in [0,16)
//        (from cdna_hip_programming.md §3: col=lane&31,
//         row=(reg&3)+8*(reg>>2)+4*(lane>>5))

#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;

__device__ inline float bf2f(__bf16 h) { return static_cast<float>(h); }

__global__ void probe(const __bf16* A, const __bf16* B, float* D) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  for (int i = 0; i < 8; ++i) {
    a[i] = A[(l % 32) * 16 + (8 * (l / 32) + i)];
    b[i] = B[(8 * (l / 32) + i) * 32 + (l % 32)];
  }
  f32x16 c{};
  for (int i = 0; i < 16; ++i) c[i] = 0.0f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg % 4) + 8 * (reg / 4) + 4 * (l / 32);
    const int col = l % 32;
    D[row * 32 + col] = c[reg];
  }
}

int main() {
  std::vector<float> Ah(32 * 16), Bh(16 * 32);
  for (int i = 0; i < 32; ++i)
    for (int k = 0; k < 16; ++k) Ah[i * 16 + k] = 0.01f * i + 0.5f * k;
  for (int k = 0; k < 16; ++k)
    for (int j = 0; j < 32; ++j) Bh[k * 32 + j] = 0.02f * j - 0.3f * k;
  std::vector<__bf16> Ab(32 * 16), Bb(16 * 32);
  for (int i = 0; i < 32 * 16; ++i) Ab[i] = static_cast<__bf16>(Ah[i]);
  for (int i = 0; i < 16 * 32; ++i) Bb[i] = static_cast<__bf16>(Bh[i]);

  __bf16 *dA, *dB;
  float* dD;
  hipMalloc(&dA, sizeof(__bf16) * 32 * 16);
  hipMalloc(&dB, sizeof(__bf16) * 16 * 32);
  hipMalloc(&dD, sizeof(float) * 32 * 32);
  hipMemcpy(dA, Ab.data(), sizeof(__bf16) * 32 * 16, hipMemcpyHostToDevice);
  hipMemcpy(dB, Bb.data(), sizeof(__bf16) * 16 * 32, hipMemcpyHostToDevice);
  probe<<<1, 64>>>(dA, dB, dD);
  hipDeviceSynchronize();
  std::vector<float> Dh(32 * 32);
  hipMemcpy(Dh.data(), dD, sizeof(float) * 32 * 32, hipMemcpyDeviceToHost);

  int bad = 0;
  float max_err = 0.0f;
  for (int i = 0; i < 32; ++i) {
    for (int j = 0; j < 32; ++j) {
      float ref = 0.0f;
      for (int k = 0; k < 16; ++k) {
        // bf16-rounded inputs for a fair reference
        ref += static_cast<float>(static_cast<__bf16>(Ah[i * 16 + k])) *
               static_cast<float>(static_cast<__bf16>(Bh[k * 32 + j]));
      }
      const float got = Dh[i * 32 + j];
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
