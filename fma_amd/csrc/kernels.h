// Shared declarations between the HIP kernels (kernels.hip) and the
// torch-extension host code (actuator.cpp).
#pragma once

#include <hip/hip_runtime.h>

// One copy descriptor: both pointers must be 16-byte aligned (enforced by
// the host side; device allocations are 256-B aligned, arena offsets are
// padded to FMA_ARENA_ALIGN).
struct FmaCopyDesc {
  const unsigned char* src;
  unsigned char* dst;
  unsigned long long bytes;
};

// LDS-staged prefix + descriptor tables bound the descriptor count per
// launch: (n+1)*8 + n*24 B of dynamic LDS, kept under the default 64 KiB
// dynamic-LDS-per-workgroup limit. The chunked planner keeps per-chunk
// descriptor counts far below this (raise chunk_bytes if a model with
// thousands of tiny tensors ever trips it).
#define FMA_MAX_DESCS_PER_LAUNCH 2000
#define FMA_ARENA_ALIGN 256

extern "C" hipError_t fma_launch_batched_copy(const FmaCopyDesc* descs_dev,
                                              const unsigned long long* prefix_dev,
                                              int ndesc,
                                              unsigned long long total_units,
                                              hipStream_t stream);

extern "C" hipError_t fma_launch_contiguous_copy(const void* src, void* dst,
                                                 unsigned long long bytes,
                                                 hipStream_t stream);

extern "C" hipError_t fma_launch_cache_invalidate(hipStream_t stream);

extern "C" hipError_t fma_launch_gemv_bf16(const void* W, const void* x,
                                           float* y, int M, int K,
                                           hipStream_t stream);

extern "C" hipError_t fma_launch_gemv_multi_bf16(
    const void* W0, int M0, void* y0, const void* W1, int M1, void* y1,
    const void* W2, int M2, void* y2, const void* x, int K,
    const void* norm_w, float norm_eps, const void* b0, const void* b1,
    const void* b2, hipStream_t stream);

extern "C" hipError_t fma_launch_gemv_bf16_out16(const void* W, const void* x,
                                                 void* y, const void* residual,
                                                 int M, int K,
                                                 hipStream_t stream);
extern "C" hipError_t fma_launch_gemv_silu_bf16_out16(
    const void* W, const void* gate, const void* up, void* y,
    const void* residual, int M, int K, hipStream_t stream);

extern "C" hipError_t fma_launch_rmsnorm1_bf16(const void* x, const void* w,
                                               void* y, int H, float eps,
                                               hipStream_t stream);
extern "C" hipError_t fma_launch_silu_mul_bf16(const void* g, const void* u,
                                               void* y, int N,
                                               hipStream_t stream);
extern "C" hipError_t fma_launch_rope_qkv_store_bf16(
    void* q, const void* k, const void* v, void* kcache, void* vcache,
    const float* cos_tab, const float* sin_tab, const int* pos_dev,
    int pos, int q_heads, int kv_heads, int half_hd, hipStream_t stream);

extern "C" hipError_t fma_launch_rope1_bf16(void* q, const float* cos_row,
                                            const float* sin_row, int heads,
                                            int half_hd, hipStream_t stream);

extern "C" int fma_attn_decode_chunks(int t, int q_heads);
extern "C" hipError_t fma_launch_attn_decode_bf16(
    const void* q, const void* K, const void* V, void* out, int t,
    int q_heads, int kv_heads, int hd, long long k_stride,
    float* partials, int chunks, const int* t_dev, hipStream_t stream);

extern "C" int fma_attn_prefill_chunks(int T, int pos0, int q_heads);

extern "C" hipError_t fma_launch_attn_prefill_bf16(
    const void* Q, const void* K, const void* V, void* O, int T, int pos0,
    int q_heads, int kv_heads, int hd, float* partials, int chunks,
    hipStream_t stream);

// 16-row MFMA prefill variant (v_mfma_f32_16x16x32_bf16); single-chunk
// only, opt-in via FMA_PREFILL_16=1
extern "C" hipError_t fma_launch_attn_prefill16_bf16(
    const void* Q, const void* K, const void* V, void* O, int T, int pos0,
    int q_heads, int kv_heads, int hd, hipStream_t stream);
