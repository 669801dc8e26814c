// CDNA4 (gfx950) kernels for the FMA sleep/wake hot path.
//
// The job: move every parameter shard of a model between its (scattered)
// device storages and one contiguous buffer — the "pack" direction feeds a
// single D2H transfer into pinned host DRAM on sleep(level=1), the "scatter"
// direction re-materializes tensors after the H2D transfer on wake_up.
// (Native replacement for what the reference delegates to vLLM's
// CuMemAllocator offload; reference README.md:16-26,
// pkg/controller/dual-pods/inference-server.go:1497,1712.)
//
// Design notes (MI355X):
// - Pure streaming copy => memory-bound. 16 B/lane loads/stores (uint4) give
//   1 KiB per wave-instruction, the coalescing sweet spot; HBM3E ceiling is
//   ~6.3 TB/s measured (MI355X_MICROARCH.md §HBM).
// - One flat global index space of 16-byte units across all descriptors;
//   each thread binary-searches the per-descriptor prefix table (staged in
//   LDS, ~10 steps for <=4k tensors) and copies one unit per grid-stride
//   iteration. Tails (<16 B) handled by the unit that owns them.
// - The same kernel serves D2D (staging), D2H and H2D with pinned host
//   memory mapped into the device address space — whichever wins on the
//   hardware is chosen by the host-side pipeline (actuator.cpp).
// - Grid sized >> 256 workgroups to fill all 8 XCDs; plain linear
//   blockIdx->unit mapping is already XCD-interleaved for streaming.

#include <hip/hip_runtime.h>

#include <cstdlib>

#include "kernels.h"

#ifndef FMA_COPY_BLOCK
#define FMA_COPY_BLOCK 256
#endif

namespace {

__global__ __launch_bounds__(FMA_COPY_BLOCK) void batched_copy_kernel(
    const FmaCopyDesc* __restrict__ descs,
    const unsigned long long* __restrict__ prefix,  // ndesc+1 entries, 16B units
    int ndesc,
    unsigned long long total_units) {
  // Stage descriptors + prefix in LDS once per block.
  extern __shared__ unsigned char s_mem[];
  unsigned long long* s_prefix = reinterpret_cast<unsigned long long*>(s_mem);
  FmaCopyDesc* s_descs =
      reinterpret_cast<FmaCopyDesc*>(s_mem + (ndesc + 1) * sizeof(unsigned long long));
  for (int i = threadIdx.x; i <= ndesc; i += blockDim.x) {
    s_prefix[i] = prefix[i];
    if (i < ndesc) s_descs[i] = descs[i];
  }
  __syncthreads();

  // Each block owns one CONTIGUOUS span of the flat unit space, walked in
  // blockDim-sized rows (consecutive lanes -> consecutive 16-B units, fully
  // coalesced). Because a thread's units are monotonically increasing and
  // a span only overlaps a handful of descriptors, the descriptor lookup
  // is one binary search at span entry plus an amortized O(1) forward
  // advance per row — a per-unit binary search (a serial ~8-step LDS
  // dependent chain) measured only ~450 GB/s on gfx950; this form is
  // memory-bound.
  const unsigned long long span =
      (total_units + gridDim.x - 1) / gridDim.x;
  const unsigned long long begin =
      static_cast<unsigned long long>(blockIdx.x) * span;
  if (begin >= total_units) return;
  const unsigned long long end = min(begin + span, total_units);

  unsigned long long u = begin + threadIdx.x;
  // binary search once: largest d with prefix[d] <= u (clamped)
  int lo = 0;
  {
    const unsigned long long u0 = min(u, total_units - 1);
    int hi = ndesc;
    while (hi - lo > 1) {
      const int mid = (lo + hi) >> 1;
      if (s_prefix[mid] <= u0) {
        lo = mid;
      } else {
        hi = mid;
      }
    }
  }
  for (; u < end; u += blockDim.x) {
    while (s_prefix[lo + 1] <= u) ++lo;  // monotone advance, ~0-1 steps
    const FmaCopyDesc d = s_descs[lo];
    const unsigned long long byte_off = (u - s_prefix[lo]) * 16ull;
    const unsigned char* __restrict__ s = d.src + byte_off;
    unsigned char* __restrict__ t = d.dst + byte_off;
    const unsigned long long rem = d.bytes - byte_off;
    if (rem >= 16ull) {
      *reinterpret_cast<uint4*>(t) = *reinterpret_cast<const uint4*>(s);
    } else {
      for (unsigned long long i = 0; i < rem; ++i) {
        t[i] = s[i];
      }
    }
  }
}

// Simple contiguous copy at max vector width; used for arena <-> staging
// slices where no descriptor table is needed (both sides contiguous).
__global__ __launch_bounds__(FMA_COPY_BLOCK) void contiguous_copy_kernel(
    const unsigned char* __restrict__ src,
    unsigned char* __restrict__ dst,
    unsigned long long bytes) {
  const unsigned long long total_units = bytes >> 4;
  const unsigned long long stride =
      static_cast<unsigned long long>(gridDim.x) * blockDim.x;
  unsigned long long u =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;
  for (; u < total_units; u += stride) {
    reinterpret_cast<uint4*>(dst)[u] = reinterpret_cast<const uint4*>(src)[u];
  }
  // tail bytes by thread 0 of block 0
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (unsigned long long i = total_units << 4; i < bytes; ++i) {
      dst[i] = src[i];
    }
  }
}

inline int copy_grid(unsigned long long total_units) {
  // >=2048 workgroups fills 256 CUs at 8 blocks/CU; cap so tiny copies
  // don't pay launch-size overhead.
  unsigned long long blocks = (total_units + FMA_COPY_BLOCK - 1) / FMA_COPY_BLOCK;
  if (blocks > 4096ull) blocks = 4096ull;
  if (blocks == 0ull) blocks = 1ull;
  return static_cast<int>(blocks);
}

}  // namespace

extern "C" hipError_t fma_launch_batched_copy(const FmaCopyDesc* descs_dev,
                                              const unsigned long long* prefix_dev,
                                              int ndesc,
                                              unsigned long long total_units,
                                              hipStream_t stream) {
  if (ndesc <= 0 || total_units == 0) return hipSuccess;
  if (ndesc > FMA_MAX_DESCS_PER_LAUNCH) return hipErrorInvalidValue;
  const size_t lds = static_cast<size_t>(ndesc + 1) * sizeof(unsigned long long) +
      static_cast<size_t>(ndesc) * sizeof(FmaCopyDesc);
  batched_copy_kernel<<<copy_grid(total_units), FMA_COPY_BLOCK, lds, stream>>>(
      descs_dev, prefix_dev, ndesc, total_units);
  return hipGetLastError();
}

extern "C" hipError_t fma_launch_contiguous_copy(const void* src, void* dst,
                                                 unsigned long long bytes,
                                                 hipStream_t stream) {
  if (bytes == 0) return hipSuccess;
  contiguous_copy_kernel<<<copy_grid(bytes >> 4), FMA_COPY_BLOCK, 0, stream>>>(
      static_cast<const unsigned char*>(src), static_cast<unsigned char*>(dst),
      bytes);
  return hipGetLastError();
}

namespace {

// One workgroup per CU issuing an agent-scope acquire: lowers to
// `buffer_inv sc1`, invalidating that CU's vector L1 (MI355X_MICROARCH.md
// §Workgroup dispatch table). Used after a VMM unmap/remap cycle so no CU
// can serve reads at the re-mapped VA from lines cached before the remap.
__global__ void cache_invalidate_kernel() {
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
}

}  // namespace

extern "C" hipError_t fma_launch_cache_invalidate(hipStream_t stream) {
  // 2048 workgroups of one wave: >=1 lands on every one of the 256 CUs.
  cache_invalidate_kernel<<<2048, 64, 0, stream>>>();
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// Batch-1 bf16 GEMV: y[M] = W[M,K] @ x[K]  (row-major W, fp32 accumulate)
//
// Decode is weight-bandwidth-bound: every token reads all of W once. The
// shape is memory-streaming, not MFMA-shaped (K-dot per row, no reuse), so
// the kernel is a coalesced row sweep: each wave owns rows (one wave = one
// row per iteration), lanes read 16 B of W per step (8 bf16), x is staged
// once per block in LDS, partials reduce across the wave. hipBLASLt's
// batch-1 GEMV measured well below link rate on these shapes (~1 TB/s
// effective end-to-end decode); this kernel targets the streaming ceiling.
// ---------------------------------------------------------------------------

namespace {

__device__ inline float bf16_to_f32(unsigned short h) {
  union {
    unsigned int u;
    float f;
  } c;
  c.u = static_cast<unsigned int>(h) << 16;
  return c.f;
}

__device__ inline unsigned short f32_to_bf16(float f) {
  union {
    float f;
    unsigned int u;
  } c;
  c.f = f;
  if ((c.u & 0x7fffffffu) > 0x7f800000u) {  // NaN: quiet, torch-compatible
    return static_cast<unsigned short>((c.u >> 16) | 0x0040u);
  }
  const unsigned int lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;  // round to nearest even
  return static_cast<unsigned short>(c.u >> 16);
}


__device__ inline void gemv_store(float* y, int row, float acc) {
  y[row] = acc;
}
__device__ inline void gemv_store(unsigned short* y, int row, float acc) {
  y[row] = f32_to_bf16(acc);
}

// kSilu: x is given as the PAIR (g, u); the GEMV input is
// silu(g)*u, computed during the LDS stage (or on the fly when x does
// not fit LDS) with exactly silu_mul_bf16_kernel's fp32 math — the
// separate activation launch disappears.
template <bool kUseLds, typename OutT, bool kSilu = false>
__global__ __launch_bounds__(256) void gemv_bf16_kernel(
    const unsigned short* __restrict__ W,  // [M, K] row-major bf16
    const unsigned short* __restrict__ x,  // [K] bf16 (kSilu: gate)
    OutT* __restrict__ y,                  // [M] fp32 or bf16
    const unsigned short* __restrict__ r,  // optional residual [M] bf16
    int M, int K,
    const unsigned short* __restrict__ x2 = nullptr) {  // kSilu: up
  // x staged in LDS when it fits without hurting occupancy; for wide K
  // (w_down shapes) every wave reads the same x slices, which the L2
  // broadcasts — 57 KB of LDS would cap residency at 2 blocks/CU and
  // measured 3.1 TB/s vs ~6 with this split.
  extern __shared__ unsigned short s_x[];
  const unsigned short* xsrc = x;
  if (kUseLds) {
    for (int i = threadIdx.x; i * 8 < K; i += blockDim.x) {
      if (kSilu) {
        const uint4 gv = reinterpret_cast<const uint4*>(x)[i];
        const uint4 uv = reinterpret_cast<const uint4*>(x2)[i];
        const unsigned short* gh =
            reinterpret_cast<const unsigned short*>(&gv);
        const unsigned short* uh =
            reinterpret_cast<const unsigned short*>(&uv);
        uint4 ov;
        unsigned short* oh = reinterpret_cast<unsigned short*>(&ov);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float gf = bf16_to_f32(gh[j]);
          const float sg = gf / (1.0f + __expf(-gf));
          oh[j] = f32_to_bf16(sg * bf16_to_f32(uh[j]));
        }
        reinterpret_cast<uint4*>(s_x)[i] = ov;
      } else {
        reinterpret_cast<uint4*>(s_x)[i] =
            reinterpret_cast<const uint4*>(x)[i];
      }
    }
    __syncthreads();
    xsrc = s_x;
  }

  const int lane = threadIdx.x & 63;
  const int wave_in_block = threadIdx.x >> 6;
  const int waves_per_block = blockDim.x >> 6;
  const int global_wave = blockIdx.x * waves_per_block + wave_in_block;
  const int total_waves = gridDim.x * waves_per_block;
  const int vec_k = K >> 3;  // uint4 (8 bf16) elements per row

  for (int row = global_wave; row < M; row += total_waves) {
    const uint4* wrow = reinterpret_cast<const uint4*>(W) +
                        static_cast<long long>(row) * vec_k;
    float acc = 0.0f;
    for (int i = lane; i < vec_k; i += 64) {
      const uint4 wv = wrow[i];
      const uint4 xv = reinterpret_cast<const uint4*>(xsrc)[i];
      const unsigned short* wh = reinterpret_cast<const unsigned short*>(&wv);
      const unsigned short* xh = reinterpret_cast<const unsigned short*>(&xv);
      if (kSilu && !kUseLds) {
        const uint4 uv = reinterpret_cast<const uint4*>(x2)[i];
        const unsigned short* uh =
            reinterpret_cast<const unsigned short*>(&uv);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float gf = bf16_to_f32(xh[j]);
          const float sg = gf / (1.0f + __expf(-gf));
          const float xv2 = bf16_to_f32(
              f32_to_bf16(sg * bf16_to_f32(uh[j])));
          acc = fmaf(bf16_to_f32(wh[j]), xv2, acc);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          acc = fmaf(bf16_to_f32(wh[j]), bf16_to_f32(xh[j]), acc);
        }
      }
    }
    // wave-wide reduction
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      acc += __shfl_down(acc, off, 64);
    }
    if (lane == 0) {
      if (r != nullptr) acc += bf16_to_f32(r[row]);
      gemv_store(y, row, acc);
    }
  }
}

}  // namespace

extern "C" hipError_t fma_launch_gemv_bf16(const void* W, const void* x,
                                           float* y, int M, int K,
                                           hipStream_t stream) {
  if ((K & 7) != 0) return hipErrorInvalidValue;
  const int block = 256;
  const int waves_per_block = block / 64;
  // >= 2048 waves fills the chip; cap blocks at the row count
  int blocks = (M + waves_per_block - 1) / waves_per_block;
  if (blocks > 2048) blocks = 2048;
  const size_t lds = static_cast<size_t>(K) * sizeof(unsigned short);
  if (lds <= 32 * 1024) {  // >= 5 blocks/CU with x staged
    gemv_bf16_kernel<true, float><<<blocks, block, lds, stream>>>(
        static_cast<const unsigned short*>(W),
        static_cast<const unsigned short*>(x), y, nullptr, M, K);
  } else {
    gemv_bf16_kernel<false, float><<<blocks, block, 0, stream>>>(
        static_cast<const unsigned short*>(W),
        static_cast<const unsigned short*>(x), y, nullptr, M, K);
  }
  return hipGetLastError();
}

// Up to 3 GEMVs sharing one x in a single launch (qkv projections,
// gate+up): one virtual row space M0+M1+M2, per-row segment lookup.
// Kills 2 launch gaps per use and fills the chip even when the small
// KV projections (M=1024 rows) alone could not.
// kNorm folds the rmsnorm producing x into this launch: every block
// stages RAW x, computes sum-of-squares with EXACTLY rmsnorm1's
// accumulation pattern (bit-identical inv_rms), then normalizes its
// staged copy in place — one launch instead of rmsnorm + gemv, and the
// norm result never round-trips HBM. Requires the LDS-staged path.
template <bool kUseLds, bool kNorm = false>
__global__ __launch_bounds__(256) void gemv_multi_bf16_kernel(
    const unsigned short* __restrict__ W0, int M0,
    unsigned short* __restrict__ y0,
    const unsigned short* __restrict__ W1, int M1,
    unsigned short* __restrict__ y1,
    const unsigned short* __restrict__ W2, int M2,
    unsigned short* __restrict__ y2,
    const unsigned short* __restrict__ x, int K,
    const unsigned short* __restrict__ norm_w, float norm_eps,
    const unsigned short* __restrict__ b0 = nullptr,  // optional biases
    const unsigned short* __restrict__ b1 = nullptr,
    const unsigned short* __restrict__ b2 = nullptr) {
  static_assert(!kNorm || kUseLds, "norm fusion needs the LDS path");
  extern __shared__ unsigned short s_x[];
  __shared__ float s_red[256];
  const unsigned short* xsrc = x;
  if (kUseLds) {
    for (int i = threadIdx.x; i * 8 < K; i += blockDim.x) {
      reinterpret_cast<uint4*>(s_x)[i] =
          reinterpret_cast<const uint4*>(x)[i];
    }
    __syncthreads();
    xsrc = s_x;
  }
  if (kNorm) {
    // sum of squares: same per-thread element mapping + tree as
    // rmsnorm1_bf16_kernel so inv_rms is bit-identical to the unfused
    // two-kernel sequence
    float ss = 0.0f;
    const int vec_h = K >> 3;
    for (int i = threadIdx.x; i < vec_h; i += blockDim.x) {
      const uint4 xv = reinterpret_cast<const uint4*>(x)[i];
      const unsigned short* xh =
          reinterpret_cast<const unsigned short*>(&xv);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float v = bf16_to_f32(xh[j]);
        ss = fmaf(v, v, ss);
      }
    }
    s_red[threadIdx.x] = ss;
    __syncthreads();
    for (int r = 128; r > 0; r >>= 1) {
      if (threadIdx.x < r) s_red[threadIdx.x] += s_red[threadIdx.x + r];
      __syncthreads();
    }
    const float inv_rms =
        rsqrtf(s_red[0] / static_cast<float>(K) + norm_eps);
    for (int i = threadIdx.x; i < K; i += blockDim.x) {
      s_x[i] = f32_to_bf16(bf16_to_f32(s_x[i]) * inv_rms *
                           bf16_to_f32(norm_w[i]));
    }
    __syncthreads();
  }
  const int lane = threadIdx.x & 63;
  const int wave_in_block = threadIdx.x >> 6;
  const int waves_per_block = blockDim.x >> 6;
  const int global_wave = blockIdx.x * waves_per_block + wave_in_block;
  const int total_waves = gridDim.x * waves_per_block;
  const int vec_k = K >> 3;
  const int M = M0 + M1 + M2;
  for (int row = global_wave; row < M; row += total_waves) {
    const unsigned short* W;
    const unsigned short* bias;
    unsigned short* y;
    int r;
    if (row < M0) {
      W = W0; y = y0; bias = b0; r = row;
    } else if (row < M0 + M1) {
      W = W1; y = y1; bias = b1; r = row - M0;
    } else {
      W = W2; y = y2; bias = b2; r = row - M0 - M1;
    }
    const uint4* wrow = reinterpret_cast<const uint4*>(W) +
                        static_cast<long long>(r) * vec_k;
    float acc = 0.0f;
    for (int i = lane; i < vec_k; i += 64) {
      const uint4 wv = wrow[i];
      const uint4 xv = reinterpret_cast<const uint4*>(xsrc)[i];
      const unsigned short* wh = reinterpret_cast<const unsigned short*>(&wv);
      const unsigned short* xh = reinterpret_cast<const unsigned short*>(&xv);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        acc = fmaf(bf16_to_f32(wh[j]), bf16_to_f32(xh[j]), acc);
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      acc += __shfl_down(acc, off, 64);
    }
    if (lane == 0) {
      if (bias != nullptr) acc += bf16_to_f32(bias[r]);
      gemv_store(y, r, acc);
    }
  }
}

extern "C" hipError_t fma_launch_gemv_multi_bf16(
    const void* W0, int M0, void* y0, const void* W1, int M1, void* y1,
    const void* W2, int M2, void* y2, const void* x, int K,
    const void* norm_w, float norm_eps, const void* b0, const void* b1,
    const void* b2, hipStream_t stream) {
  if ((K & 7) != 0) return hipErrorInvalidValue;
  const int block = 256;
  const int waves_per_block = block / 64;
  const int M = M0 + M1 + M2;
  int blocks = (M + waves_per_block - 1) / waves_per_block;
  if (blocks > 2048) blocks = 2048;
  const size_t lds = static_cast<size_t>(K) * sizeof(unsigned short);
  if (norm_w != nullptr) {
    if (lds > 32 * 1024) return hipErrorInvalidValue;
    gemv_multi_bf16_kernel<true, true><<<blocks, block, lds, stream>>>(
        static_cast<const unsigned short*>(W0), M0,
        static_cast<unsigned short*>(y0),
        static_cast<const unsigned short*>(W1), M1,
        static_cast<unsigned short*>(y1),
        static_cast<const unsigned short*>(W2), M2,
        static_cast<unsigned short*>(y2),
        static_cast<const unsigned short*>(x), K,
        static_cast<const unsigned short*>(norm_w), norm_eps,
        static_cast<const unsigned short*>(b0),
        static_cast<const unsigned short*>(b1),
        static_cast<const unsigned short*>(b2));
    return hipGetLastError();
  }
#define FMA_GEMVM_ARGS                                                    static_cast<const unsigned short*>(W0), M0,                                 static_cast<unsigned short*>(y0),                                       static_cast<const unsigned short*>(W1), M1,                             static_cast<unsigned short*>(y1),                                       static_cast<const unsigned short*>(W2), M2,                             static_cast<unsigned short*>(y2),                                       static_cast<const unsigned short*>(x), K
  if (lds <= 32 * 1024) {
    gemv_multi_bf16_kernel<true><<<blocks, block, lds, stream>>>(
        FMA_GEMVM_ARGS, nullptr, 0.0f,
        static_cast<const unsigned short*>(b0),
        static_cast<const unsigned short*>(b1),
        static_cast<const unsigned short*>(b2));
  } else {
    gemv_multi_bf16_kernel<false><<<blocks, block, 0, stream>>>(
        FMA_GEMVM_ARGS, nullptr, 0.0f,
        static_cast<const unsigned short*>(b0),
        static_cast<const unsigned short*>(b1),
        static_cast<const unsigned short*>(b2));
  }
#undef FMA_GEMVM_ARGS
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// Fused decode elementwise kernels. Profiling the eager decode showed more
// GPU time in fragmented elementwise kernels (rmsnorm = 5 launches, rope,
// silu*up, bf16 converts) than in the GEMVs that do the actual weight
// reads; each fused op here is one launch, fp32 math, RNE bf16 out.
// ---------------------------------------------------------------------------

namespace {

// One block: y = x * rsqrt(mean(x^2) + eps) * w  (single token, H <= 64k)
__global__ __launch_bounds__(256) void rmsnorm1_bf16_kernel(
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ w,
    unsigned short* __restrict__ y, int H, float eps) {
  __shared__ float red[256];
  float ss = 0.0f;
  const int vec_h = H >> 3;
  for (int i = threadIdx.x; i < vec_h; i += blockDim.x) {
    const uint4 xv = reinterpret_cast<const uint4*>(x)[i];
    const unsigned short* xh = reinterpret_cast<const unsigned short*>(&xv);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float v = bf16_to_f32(xh[j]);
      ss = fmaf(v, v, ss);
    }
  }
  red[threadIdx.x] = ss;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (threadIdx.x < s) red[threadIdx.x] += red[threadIdx.x + s];
    __syncthreads();
  }
  const float scale = rsqrtf(red[0] / H + eps);
  for (int i = threadIdx.x; i < vec_h; i += blockDim.x) {
    const uint4 xv = reinterpret_cast<const uint4*>(x)[i];
    const uint4 wv = reinterpret_cast<const uint4*>(w)[i];
    const unsigned short* xh = reinterpret_cast<const unsigned short*>(&xv);
    const unsigned short* wh = reinterpret_cast<const unsigned short*>(&wv);
    uint4 ov;
    unsigned short* oh = reinterpret_cast<unsigned short*>(&ov);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      oh[j] = f32_to_bf16(bf16_to_f32(xh[j]) * scale * bf16_to_f32(wh[j]));
    }
    reinterpret_cast<uint4*>(y)[i] = ov;
  }
}

// y = silu(g) * u, elementwise over N bf16 values (N % 8 == 0)
__global__ __launch_bounds__(256) void silu_mul_bf16_kernel(
    const unsigned short* __restrict__ g,
    const unsigned short* __restrict__ u,
    unsigned short* __restrict__ y, int N) {
  const int vec_n = N >> 3;
  const int stride = gridDim.x * blockDim.x;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < vec_n; i += stride) {
    const uint4 gv = reinterpret_cast<const uint4*>(g)[i];
    const uint4 uv = reinterpret_cast<const uint4*>(u)[i];
    const unsigned short* gh = reinterpret_cast<const unsigned short*>(&gv);
    const unsigned short* uh = reinterpret_cast<const unsigned short*>(&uv);
    uint4 ov;
    unsigned short* oh = reinterpret_cast<unsigned short*>(&ov);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf16_to_f32(gh[j]);
      const float s = gf / (1.0f + __expf(-gf));
      oh[j] = f32_to_bf16(s * bf16_to_f32(uh[j]));
    }
    reinterpret_cast<uint4*>(y)[i] = ov;
  }
}

// In-place RoPE on one token: q [heads, hd] as (x0,x1) pairs; cos/sin [hd/2]
// fp32 (one row of the host-precomputed table).
__global__ __launch_bounds__(256) void rope1_bf16_kernel(
    unsigned short* __restrict__ q, const float* __restrict__ cos_row,
    const float* __restrict__ sin_row, int heads, int half_hd) {
  const int total = heads * half_hd;
  const int stride = gridDim.x * blockDim.x;
  for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < total; p += stride) {
    const int d = p % half_hd;
    const float c = cos_row[d];
    const float s = sin_row[d];
    const float x0 = bf16_to_f32(q[2 * p]);
    const float x1 = bf16_to_f32(q[2 * p + 1]);
    q[2 * p] = f32_to_bf16(x0 * c - x1 * s);
    q[2 * p + 1] = f32_to_bf16(x0 * s + x1 * c);
  }
}

}  // namespace

extern "C" hipError_t fma_launch_gemv_bf16_out16(const void* W, const void* x,
                                                 void* y, const void* residual,
                                                 int M, int K,
                                                 hipStream_t stream) {
  if ((K & 7) != 0) return hipErrorInvalidValue;
  const int block = 256;
  const int waves_per_block = block / 64;
  int blocks = (M + waves_per_block - 1) / waves_per_block;
  if (blocks > 2048) blocks = 2048;
  const size_t lds = static_cast<size_t>(K) * sizeof(unsigned short);
  if (lds <= 32 * 1024) {
    gemv_bf16_kernel<true, unsigned short><<<blocks, block, lds, stream>>>(
        static_cast<const unsigned short*>(W),
        static_cast<const unsigned short*>(x),
        static_cast<unsigned short*>(y),
        static_cast<const unsigned short*>(residual), M, K);
  } else {
    gemv_bf16_kernel<false, unsigned short><<<blocks, block, 0, stream>>>(
        static_cast<const unsigned short*>(W),
        static_cast<const unsigned short*>(x),
        static_cast<unsigned short*>(y),
        static_cast<const unsigned short*>(residual), M, K);
  }
  return hipGetLastError();
}

extern "C" hipError_t fma_launch_gemv_silu_bf16_out16(
    const void* W, const void* gate, const void* up, void* y,
    const void* residual, int M, int K, hipStream_t stream) {
  if ((K & 7) != 0) return hipErrorInvalidValue;
  const int block = 256;
  const int waves_per_block = block / 64;
  int blocks = (M + waves_per_block - 1) / waves_per_block;
  if (blocks > 2048) blocks = 2048;
  const size_t lds = static_cast<size_t>(K) * sizeof(unsigned short);
  if (lds <= 32 * 1024) {
    gemv_bf16_kernel<true, unsigned short, true>
        <<<blocks, block, lds, stream>>>(
            static_cast<const unsigned short*>(W),
            static_cast<const unsigned short*>(gate),
            static_cast<unsigned short*>(y),
            static_cast<const unsigned short*>(residual), M, K,
            static_cast<const unsigned short*>(up));
  } else {
    gemv_bf16_kernel<false, unsigned short, true>
        <<<blocks, block, 0, stream>>>(
            static_cast<const unsigned short*>(W),
            static_cast<const unsigned short*>(gate),
            static_cast<unsigned short*>(y),
            static_cast<const unsigned short*>(residual), M, K,
            static_cast<const unsigned short*>(up));
  }
  return hipGetLastError();
}

extern "C" hipError_t fma_launch_rmsnorm1_bf16(const void* x, const void* w,
                                               void* y, int H, float eps,
                                               hipStream_t stream) {
  if ((H & 7) != 0) return hipErrorInvalidValue;
  rmsnorm1_bf16_kernel<<<1, 256, 0, stream>>>(
      static_cast<const unsigned short*>(x),
      static_cast<const unsigned short*>(w),
      static_cast<unsigned short*>(y), H, eps);
  return hipGetLastError();
}

extern "C" hipError_t fma_launch_silu_mul_bf16(const void* g, const void* u,
                                               void* y, int N,
                                               hipStream_t stream) {
  if ((N & 7) != 0) return hipErrorInvalidValue;
  int blocks = ((N >> 3) + 255) / 256;
  if (blocks > 1024) blocks = 1024;
  silu_mul_bf16_kernel<<<blocks, 256, 0, stream>>>(
      static_cast<const unsigned short*>(g),
      static_cast<const unsigned short*>(u),
      static_cast<unsigned short*>(y), N);
  return hipGetLastError();
}

extern "C" hipError_t fma_launch_rope1_bf16(void* q, const float* cos_row,
                                            const float* sin_row, int heads,
                                            int half_hd, hipStream_t stream) {
  const int total = heads * half_hd;
  int blocks = (total + 255) / 256;
  if (blocks > 1024) blocks = 1024;
  rope1_bf16_kernel<<<blocks, 256, 0, stream>>>(
      static_cast<unsigned short*>(q), cos_row, sin_row, heads, half_hd);
  return hipGetLastError();
}

namespace {

// One launch for the whole decode attention pre-step: RoPE(q) in place,
// RoPE(k) written into its KV-cache row, v copied into its row — folds
// 4 tiny launch-bound kernels (2 rope + 2 index_copy) into one. The
// position (cos/sin table row + cache row) is read from a device int32
// so a captured hipGraph replays correctly as the cache grows.
__global__ __launch_bounds__(256) void rope_qkv_store_bf16_kernel(
    unsigned short* __restrict__ q,        // [qH, hd], roped in place
    const unsigned short* __restrict__ k,  // [kvH, hd]
    const unsigned short* __restrict__ v,  // [kvH, hd]
    unsigned short* __restrict__ kcache,   // [S, kvH, hd] base
    unsigned short* __restrict__ vcache,
    const float* __restrict__ cos_tab,     // [S, hd/2]
    const float* __restrict__ sin_tab,
    const int* __restrict__ pos_dev,       // device position (or null)
    int pos_arg, int q_heads, int kv_heads, int half_hd) {
  const int pos = pos_dev != nullptr ? *pos_dev : pos_arg;
  const float* cos_row = cos_tab + static_cast<long long>(pos) * half_hd;
  const float* sin_row = sin_tab + static_cast<long long>(pos) * half_hd;
  const long long row = static_cast<long long>(pos) * kv_heads *
                        (2 * half_hd);
  unsigned short* kdst = kcache + row;
  unsigned short* vdst = vcache + row;
  const int qtotal = q_heads * half_hd;
  const int kvtotal = kv_heads * half_hd;
  const int stride = gridDim.x * blockDim.x;
  for (int p = blockIdx.x * blockDim.x + threadIdx.x;
       p < qtotal + 2 * kvtotal; p += stride) {
    if (p < qtotal) {
      const int d = p % half_hd;
      const float c = cos_row[d];
      const float sn = sin_row[d];
      const float x0 = bf16_to_f32(q[2 * p]);
      const float x1 = bf16_to_f32(q[2 * p + 1]);
      q[2 * p] = f32_to_bf16(x0 * c - x1 * sn);
      q[2 * p + 1] = f32_to_bf16(x0 * sn + x1 * c);
    } else if (p < qtotal + kvtotal) {
      const int pk = p - qtotal;
      const int d = pk % half_hd;
      const float c = cos_row[d];
      const float sn = sin_row[d];
      const float x0 = bf16_to_f32(k[2 * pk]);
      const float x1 = bf16_to_f32(k[2 * pk + 1]);
      kdst[2 * pk] = f32_to_bf16(x0 * c - x1 * sn);
      kdst[2 * pk + 1] = f32_to_bf16(x0 * sn + x1 * c);
    } else {
      const int pv = p - qtotal - kvtotal;
      reinterpret_cast<uint*>(vdst)[pv] =
          reinterpret_cast<const uint*>(v)[pv];
    }
  }
}

}  // namespace

extern "C" hipError_t fma_launch_rope_qkv_store_bf16(
    void* q, const void* k, const void* v, void* kcache, void* vcache,
    const float* cos_tab, const float* sin_tab, const int* pos_dev,
    int pos, int q_heads, int kv_heads, int half_hd, hipStream_t stream) {
  const int total = (q_heads + 2 * kv_heads) * half_hd;
  int blocks = (total + 255) / 256;
  if (blocks > 512) blocks = 512;
  rope_qkv_store_bf16_kernel<<<blocks, 256, 0, stream>>>(
      static_cast<unsigned short*>(q),
      static_cast<const unsigned short*>(k),
      static_cast<const unsigned short*>(v),
      static_cast<unsigned short*>(kcache),
      static_cast<unsigned short*>(vcache), cos_tab, sin_tab, pos_dev,
      pos, q_heads, kv_heads, half_hd);
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// Single-token GQA decode attention (flash-style online softmax).
//
// out[qh] = softmax(q[qh] . K[kv(qh), :t] / sqrt(hd)) @ V[kv(qh), :t]
//
// One workgroup (4 waves) per query head; each wave walks seq positions
// (4 at a time per block), lanes split head_dim (2 elements each -> the
// 256 B K/V row reads are fully coalesced and the per-wave accumulator
// is just 2 VGPRs/lane). Replaces the eager path's repeat_interleave
// copies (2 full-cache copies per layer) and ~6 launches per layer.
// Layout: K/V rows may be strided (cache is [S, kvH, hd]).
// ---------------------------------------------------------------------------

namespace {

#define FMA_ATTN_MAX_T 8192

// Split-sequence ("flash-decode") layout: grid.x = q_heads, grid.y =
// seq chunks, so long contexts fill all 256 CUs instead of q_heads
// blocks. Each block emits an (m, l, acc[hd]) partial; a tiny combine
// kernel merges chunks per head. chunks == 1 writes the output directly.
__global__ __launch_bounds__(256) void attn_decode_bf16_kernel(
    const unsigned short* __restrict__ q,   // [qH, hd]
    const unsigned short* __restrict__ K,   // rows: K + s*k_stride + kvh*hd
    const unsigned short* __restrict__ V,
    unsigned short* __restrict__ out,       // [qH, hd] (chunks == 1)
    float* __restrict__ partials,           // [qH, chunks, hd + 2]
    const int* __restrict__ t_dev,          // runtime t (hipGraph mode)|null
    int t_arg, int q_heads, int kv_heads, int hd,
    long long k_stride) {
  // hipGraph mode: the sequence length is read from device memory so a
  // captured graph replays correctly as the KV cache grows (t_arg is
  // then only the upper bound the launch geometry was sized for)
  const int t = t_dev != nullptr ? *t_dev : t_arg;
  __shared__ float s_scores[FMA_ATTN_MAX_T / 4];
  __shared__ float s_q[256];
  __shared__ float s_red[256];
  __shared__ float s_wacc16[16][256];

  // chunk on X, head on Y: consecutive blockIdx (round-robin over the 8
  // XCDs) then maps the SAME key chunk of every q-head to the SAME XCD
  // whenever gridDim.x % 8 == 0 ((c + chunks*qh) % 8 == c % 8), so the 4
  // q-heads of a GQA group hit their kv-head's K/V tile in that XCD's L2
  // instead of re-reading HBM 4x.
  const int qh = blockIdx.y;
  const int chunk = blockIdx.x;
  const int chunks = gridDim.x;
  const int span = (t + chunks - 1) / chunks;
  const int s_begin = chunk * span;
  const int s_end = min(s_begin + span, t);
  const int n = s_end - s_begin;
  const int kvh = qh / (q_heads / kv_heads);
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves = blockDim.x >> 6;
  const int per_lane = hd / 64;
  const float scale = rsqrtf(static_cast<float>(hd));

  for (int i = threadIdx.x; i < hd; i += blockDim.x) {
    s_q[i] = bf16_to_f32(q[qh * hd + i]) * scale;
  }
  __syncthreads();
  if (n <= 0) {
    if (chunks > 1 && threadIdx.x == 0) {
      float* p = partials + (static_cast<long long>(qh) * chunks + chunk) *
                                (hd + 2);
      p[hd] = -1e30f;
      p[hd + 1] = 0.0f;
    }
    if (chunks > 1) {
      for (int i = threadIdx.x; i < hd; i += blockDim.x) {
        partials[(static_cast<long long>(qh) * chunks + chunk) * (hd + 2) +
                 i] = 0.0f;
      }
    }
    return;
  }

  const unsigned short* kbase = K + static_cast<long long>(kvh) * hd;
  const unsigned short* vbase = V + static_cast<long long>(kvh) * hd;

  // pass 1: scores, wave-cooperative. One wave covers one K row with a
  // CONTIGUOUS 256 B load (lane -> hd/64 consecutive elements) and
  // reduces the dot across lanes — the old thread-per-row layout made
  // every wavefront touch 64 rows at k_stride apart, collapsing HBM
  // efficiency ~10x at long t (the 4K-decode deficit in round 1).
  // 16 lanes per row, 4 rows per wave in parallel: one contiguous
  // hd/16-element load per lane covers the row, the dot reduces in 4
  // shfl steps within the 16-lane group (vs 6 across the full wave), and
  // the four rows' loads issue under one HBM latency.
  float local_max = -1e30f;
  const int g16 = lane >> 4;        // which of the wave's 4 rows
  const int lane16 = lane & 15;     // position within the row
  const int epl = hd / 16;          // elements per lane (8 at hd=128)
  for (int r0 = wave * 4; r0 < n + 3; r0 += waves * 4) {
    const int row = r0 + g16;
    float d = 0.0f;
    if (row < n) {
      const unsigned short* krow = kbase + (s_begin + row) * k_stride;
#pragma unroll 8
      for (int j = 0; j < epl; ++j) {
        d = fmaf(s_q[lane16 * epl + j],
                 bf16_to_f32(krow[lane16 * epl + j]), d);
      }
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) {
      d += __shfl_xor(d, off, 16);
    }
    if (row < n) {
      if (lane16 == 0) s_scores[row] = d;
      local_max = fmaxf(local_max, d);
    }
  }
  s_red[threadIdx.x] = local_max;
  __syncthreads();
  for (int r = 128; r > 0; r >>= 1) {
    if (threadIdx.x < r) {
      s_red[threadIdx.x] = fmaxf(s_red[threadIdx.x], s_red[threadIdx.x + r]);
    }
    __syncthreads();
  }
  const float m = s_red[0];
  __syncthreads();
  float local_sum = 0.0f;
  for (int s0 = threadIdx.x; s0 < n; s0 += blockDim.x) {
    const float w = __expf(s_scores[s0] - m);
    s_scores[s0] = w;
    local_sum += w;
  }
  s_red[threadIdx.x] = local_sum;
  __syncthreads();
  for (int r = 128; r > 0; r >>= 1) {
    if (threadIdx.x < r) s_red[threadIdx.x] += s_red[threadIdx.x + r];
    __syncthreads();
  }
  const float l = s_red[0];

  // pass 2: weighted V accumulation with the same 16-lane row mapping
  // (one contiguous load per lane per row); the 16 row-groups' partial
  // sums merge through LDS.
  float acc[16];
#pragma unroll
  for (int j = 0; j < 16; ++j) acc[j] = 0.0f;
  for (int r0 = wave * 4; r0 < n + 3; r0 += waves * 4) {
    const int row = r0 + g16;
    if (row >= n) continue;
    const unsigned short* vrow = vbase + (s_begin + row) * k_stride;
    const float w = s_scores[row];
#pragma unroll 8
    for (int j = 0; j < epl; ++j) {
      acc[j] = fmaf(w, bf16_to_f32(vrow[lane16 * epl + j]), acc[j]);
    }
  }
#pragma unroll 8
  for (int j = 0; j < epl; ++j) {
    s_wacc16[wave * 4 + g16][lane16 * epl + j] = acc[j];
  }
  __syncthreads();
  if (wave == 0) {
    if (chunks == 1) {
      const float inv_l = 1.0f / l;
      for (int j = 0; j < per_lane; ++j) {
        float v_out = 0.0f;
#pragma unroll
        for (int g = 0; g < 16; ++g) {
          v_out += s_wacc16[g][lane * per_lane + j];
        }
        out[qh * hd + lane * per_lane + j] = f32_to_bf16(v_out * inv_l);
      }
    } else {
      float* p = partials + (static_cast<long long>(qh) * chunks + chunk) *
                                (hd + 2);
      for (int j = 0; j < per_lane; ++j) {
        float v_out = 0.0f;
#pragma unroll
        for (int g = 0; g < 16; ++g) {
          v_out += s_wacc16[g][lane * per_lane + j];
        }
        p[lane * per_lane + j] = v_out;
      }
      if (lane == 0) {
        p[hd] = m;
        p[hd + 1] = l;
      }
    }
  }
}

// Merge chunk partials: 4 waves per q head, each covering a quarter of
// the chunks (a single wave serialized 32 chunk reads and became half
// the 4K attention cost once the main kernel went wide).
__global__ __launch_bounds__(256) void attn_decode_combine_kernel(
    const float* __restrict__ partials,  // [qH, chunks, hd + 2]
    unsigned short* __restrict__ out,    // [qH, hd]
    int chunks, int hd) {
  const int qh = blockIdx.x;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int per_lane = hd / 64;
  __shared__ float s_acc[4][256];
  __shared__ float s_l[4];
  const float* base = partials + static_cast<long long>(qh) * chunks * (hd + 2);
  float m_tot = -1e30f;
  for (int c = 0; c < chunks; ++c) {
    m_tot = fmaxf(m_tot, base[c * (hd + 2) + hd]);
  }
  float l_tot = 0.0f;
  float acc[4] = {0.f, 0.f, 0.f, 0.f};
  for (int c = wave; c < chunks; c += 4) {
    const float* p = base + c * (hd + 2);
    const float alpha = __expf(p[hd] - m_tot);
    l_tot += p[hd + 1] * alpha;
    for (int j = 0; j < per_lane; ++j) {
      acc[j] += p[lane * per_lane + j] * alpha;
    }
  }
  for (int j = 0; j < per_lane; ++j) {
    s_acc[wave][lane * per_lane + j] = acc[j];
  }
  if (lane == 0) s_l[wave] = l_tot;
  __syncthreads();
  if (wave == 0) {
    const float inv =
        1.0f / (s_l[0] + s_l[1] + s_l[2] + s_l[3]);
    for (int j = 0; j < per_lane; ++j) {
      const int e = lane * per_lane + j;
      const float v = s_acc[0][e] + s_acc[1][e] + s_acc[2][e] + s_acc[3][e];
      out[qh * hd + e] = f32_to_bf16(v * inv);
    }
  }
}

}  // namespace

extern "C" int fma_attn_decode_chunks(int t, int q_heads) {
  // OVERFILL the chip: ~1024 blocks puts 4 workgroups on every CU
  // (LDS/VGPR allow it), and that occupancy is what hides the HBM
  // latency of the per-row K/V streams — at 256 blocks the kernel ran
  // at occupancy 1 and ~2% of peak at t=4K (profiles/decode4k_kernel_stats_round2.csv).
  // Keep chunks >= 128 positions so the partials+combine overhead stays
  // amortized, and within the LDS score window.
  int chunks = 512 / q_heads;  // ~512 blocks: measured best (24.7 us at
                               // t=4K vs 26.4 at 1024; tools/
                               // decode_kernel_bench.py sweep)
  if (chunks < 1) chunks = 1;
  const int max_by_span = (t + 127) / 128;
  if (chunks > max_by_span) chunks = max_by_span;
  const int min_by_lds = (t + FMA_ATTN_MAX_T / 4 - 1) / (FMA_ATTN_MAX_T / 4);
  if (chunks < min_by_lds) chunks = min_by_lds;
  // multiples of 8 keep the chunk->XCD mapping head-invariant (see the
  // grid comment in the kernel): same key chunk -> same XCD L2
  if (chunks > 8 && chunks % 8) chunks += 8 - chunks % 8;
  if (const char* e = getenv("FMA_DECODE_CHUNKS")) {
    const int forced = atoi(e);
    if (forced >= 1) {
      const int lds_min =
          (t + FMA_ATTN_MAX_T / 4 - 1) / (FMA_ATTN_MAX_T / 4);
      chunks = forced < lds_min ? lds_min : forced;
    }
  }
  return chunks;
}

extern "C" hipError_t fma_launch_attn_decode_bf16(
    const void* q, const void* K, const void* V, void* out, int t,
    int q_heads, int kv_heads, int hd, long long k_stride,
    float* partials, int chunks, const int* t_dev, hipStream_t stream) {
  if (hd > 256 || (hd & 63) != 0) return hipErrorInvalidValue;
  if (q_heads % kv_heads != 0) return hipErrorInvalidValue;
  // any t: the chunk heuristic bounds each chunk's scores to the LDS
  // window (FMA_ATTN_MAX_T/4); with t_dev, t is the static upper bound
  if ((t + chunks - 1) / chunks > FMA_ATTN_MAX_T / 4)
    return hipErrorInvalidValue;
  if (chunks > 1 && partials == nullptr) return hipErrorInvalidValue;
  dim3 grid(chunks, q_heads);
  attn_decode_bf16_kernel<<<grid, 256, 0, stream>>>(
      static_cast<const unsigned short*>(q),
      static_cast<const unsigned short*>(K),
      static_cast<const unsigned short*>(V),
      static_cast<unsigned short*>(out), partials, t_dev, t, q_heads,
      kv_heads, hd, k_stride);
  if (chunks > 1) {
    attn_decode_combine_kernel<<<q_heads, 256, 0, stream>>>(
        partials, static_cast<unsigned short*>(out), chunks, hd);
  }
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// Prefill flash attention (causal, GQA) on MFMA.
//
// S = Q·K^T and O += P·V run on v_mfma_f32_32x32x16_bf16 (fragment
// mapping validated by tools/mfma_probe.hip on gfx950); softmax is the
// standard online form in fp32. One WAVE per (query head, 32-row query
// tile); key tiles of 32 stream through, V staged in LDS, P transposed
// to A-fragment layout through LDS. Correctness-first structure (no
// double-buffering yet); numerics match SDPA-class references.
//
// Layouts: Q/O [T, qH, hd] contiguous; K/V [S, kvH, hd] contiguous
// (the model's batch-1 slices). hd in {64, 128}.
// ---------------------------------------------------------------------------

namespace {

using bf16x8_t = __attribute__((ext_vector_type(8))) __bf16;
using f32x16_t = __attribute__((ext_vector_type(16))) float;

__device__ inline float wave32_max(float v) {
#pragma unroll
  for (int off = 16; off > 0; off >>= 1) {
    v = fmaxf(v, __shfl_xor(v, off, 32));
  }
  return v;
}

__device__ inline float wave32_sum(float v) {
#pragma unroll
  for (int off = 16; off > 0; off >>= 1) {
    v += __shfl_xor(v, off, 32);
  }
  return v;
}

// s_waitcnt immediate that waits lgkmcnt(0) only (vmcnt/expcnt left
// outstanding): gfx9/CDNA encoding vmcnt[3:0]|expcnt[6:4]|lgkmcnt[11:8]
// with vmcnt high bits [15:14].
#define FMA_WAIT_LGKM0 0xC07F

// Row stride of the f32 score scratch: 33 makes every per-row transpose
// access land in a distinct LDS bank (32 would put a whole row in one).
#define FMA_SROW 33

// Load one 32-key tile of K as B-fragments (8 bf16 per lane per k-step).
template <int HD>
__device__ __forceinline__ void prefill_load_k(
    const unsigned short* __restrict__ K, int kcol0, int t_kv, int kv_heads,
    int kvh, int lane32, int half, bf16x8_t (&kf)[HD / 16]) {
  const int key = kcol0 + lane32;
  const bool live = key < t_kv;
  const unsigned short* kp =
      K + (static_cast<long long>(live ? key : 0) * kv_heads + kvh) * HD +
      8 * half;
#pragma unroll
  for (int ks = 0; ks < HD / 16; ++ks) {
    if (live) {
      kf[ks] = *reinterpret_cast<const bf16x8_t*>(kp + ks * 16);
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) kf[ks][i] = static_cast<__bf16>(0.0f);
    }
  }
}

// Stage one 32-key tile of V into LDS, transposed to [hd][key] so the
// P.V B-fragment reads are contiguous b128 loads. [c8_lo, c8_hi) selects
// the 16-channel chunks this wave stages (multi-wave workgroups split
// the tile across waves — each stages HD/(16*NW) chunks).
template <int HD>
__device__ __forceinline__ void prefill_stage_v(
    const unsigned short* __restrict__ V, int kcol0, int t_kv, int kv_heads,
    int kvh, int lane32, int half, __bf16* vdst, int c8_lo, int c8_hi) {
  const int key = kcol0 + lane32;
  const bool live = key < t_kv;
  const unsigned short* vp =
      V + (static_cast<long long>(live ? key : 0) * kv_heads + kvh) * HD +
      8 * half;
  for (int c8 = c8_lo; c8 < c8_hi; ++c8) {
    bf16x8_t vv;
    if (live) {
      vv = *reinterpret_cast<const bf16x8_t*>(vp + c8 * 16);
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) vv[i] = static_cast<__bf16>(0.0f);
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      vdst[(c8 * 16 + 8 * half + i) * 32 + lane32] = vv[i];
    }
  }
}

// NW = waves per workgroup. NW=4 covers the 4 q-heads of one GQA group
// (or 4 group-mates) with ONE shared V staging pipeline: global V reads
// and LDS fill drop 4x, and at ~34 KB LDS per 4-wave WG the CU fits 4
// workgroups = 4 waves/SIMD (the 1-wave shape is LDS-bound at 2).
// PIPE (NW>1 only): software-pipelined K prefetch + double V buffer at
// occupancy 2, vs the streaming single-buffer shape that trades ILP for
// lower LDS. Both are spill-free at waves_per_eu(2); A/B via
// FMA_PREFILL_PIPE.
template <int HD, int NW, bool PIPE = false>
__global__ __launch_bounds__(64 * NW)
__attribute__((amdgpu_waves_per_eu(2)))
void attn_prefill_bf16_kernel(
    const unsigned short* __restrict__ Q,  // [T, qH, hd]
    const unsigned short* __restrict__ K,  // [S, kvH, hd]
    const unsigned short* __restrict__ V,  // [S, kvH, hd]
    unsigned short* __restrict__ O,        // [T, qH, hd]
    float* __restrict__ partials,          // [qH,tiles,chunks,32,hd+2]|null
    int T, int pos0, int q_heads, int kv_heads) {
  // Split-sequence mode (gridDim.z > 1): small prefills launch too few
  // waves to fill 1024 SIMDs (T=512 -> 512 waves), so chunk blockIdx.z
  // takes key tiles {z, z+chunks, ...} (strided keeps the causal load
  // balanced) and writes unnormalized (m, l, O) partials; the combine
  // kernel below merges them. Same recipe as attn_decode.
  constexpr int kNblk = HD / 32;
  const int wid = NW > 1 ? (threadIdx.x >> 6) : 0;
  const int qh = blockIdx.x * NW + wid;
  // schedule the HEAVIEST tiles first: under causal masking tile i does
  // i+1 key tiles of work, so launching high tiles last would leave a
  // ragged mostly-idle tail on the 1024 SIMDs
  const int tile = gridDim.y - 1 - blockIdx.y;
  const int r0 = tile * 32;
  if (r0 >= T) return;  // uniform across the WG (depends on blockIdx only)
  const int kvh = qh / (q_heads / kv_heads);
  const int l = threadIdx.x & 63;
  const int half = l >> 5;  // 0 or 1
  const int lane32 = l & 31;
  const float scale = rsqrtf(static_cast<float>(HD));

  // NW=1: P 2 KB + two V bufs 16 KB + score scratch 2.1 KB + stats
  // ~0.3 KB = 20.4 KB -> 7-8 WGs/CU (2 waves/SIMD). NW=4: the V bufs are
  // SHARED (16 KB once), per-wave scratch x4 -> ~34 KB per 4-wave WG ->
  // 4 WGs/CU = 4 waves/SIMD.
  // NW=1 keeps the software-pipelined double V buffer; NW=4 trades the
  // prefetch for occupancy (4 waves/SIMD needs <= 128 VGPRs, and the
  // prefetch's long live ranges were pushing 242) and uses ONE V buffer
  constexpr int kVBufs = (NW > 1 && !PIPE) ? 1 : 2;
  __shared__ __bf16 s_p_all[NW][32 * 32];        // P tile, A-frag source
  __shared__ __bf16 s_v[kVBufs][32 * HD];        // V [hd][key], SHARED
  __shared__ __bf16 s_s_all[NW][32 * FMA_SROW];  // score transpose scratch
  __shared__ float s_stat_all[NW][32 * 2];       // per-row (m_new, alpha)
  __shared__ float s_l_all[NW][32];              // per-row l at epilogue
  __bf16* const s_p = s_p_all[wid];
  __bf16* const s_s = s_s_all[wid];
  float* const s_stat = s_stat_all[wid];
  float* const s_l = s_l_all[wid];
  // this wave's share of the cooperative V staging
  constexpr int kC8PerWave = (HD / 16) / NW;
  const int c8_lo = wid * kC8PerWave;
  const int c8_hi = c8_lo + kC8PerWave;

  // Q fragments, kept in registers for the whole key loop.
  // A-layout: lane holds A[lane32][8*half + i] per 16-k step.
  bf16x8_t qf[HD / 16];
  const int q_row = r0 + lane32;
  const bool row_live = q_row < T;
  if (row_live) {
    const unsigned short* qp =
        Q + (static_cast<long long>(q_row) * q_heads + qh) * HD + 8 * half;
#pragma unroll
    for (int ks = 0; ks < HD / 16; ++ks) {
      bf16x8_t qv = *reinterpret_cast<const bf16x8_t*>(qp + ks * 16);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        // pre-scale Q by 1/sqrt(hd) (cheaper than scaling S)
        qf[ks][i] = static_cast<__bf16>(static_cast<float>(qv[i]) * scale);
      }
    }
  } else {
#pragma unroll
    for (int ks = 0; ks < HD / 16; ++ks) {
#pragma unroll
      for (int i = 0; i < 8; ++i) qf[ks][i] = static_cast<__bf16>(0.0f);
    }
  }

  // Per-ROW softmax stats in "transpose layout": lane l keeps the running
  // max and sum for row l%32 (both halves redundantly); the C/D register
  // layout picks them up from LDS each tile. This replaces 10 ds_bpermute
  // chains per register (160 per tile) with conflict-free LDS transposes.
  float m_row = -1e30f, l_row = 0.0f;

  f32x16_t oacc[kNblk];
#pragma unroll
  for (int b = 0; b < kNblk; ++b) {
#pragma unroll
    for (int r = 0; r < 16; ++r) oacc[b][r] = 0.0f;
  }

  const int t_kv = pos0 + T;  // causal key horizon
  const int kt_end = min((pos0 + r0 + 31) / 32 + 1, (t_kv + 31) / 32);

  // Softmax + PV for one tile whose scores sit in sacc and whose V tile
  // is staged (transposed) at vdst. Wave-synchronous: lgkm-only waits.
  auto softmax_pv = [&](f32x16_t& sacc, const __bf16* vdst, int kcol0) {
    // phase A: masked scores into the f32 transpose scratch
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r % 4) + 8 * (r / 4) + 4 * half;
      const int qpos = pos0 + r0 + row;
      if (kcol0 + lane32 > qpos || kcol0 + lane32 >= t_kv) sacc[r] = -1e30f;
      s_s[row * FMA_SROW + lane32] = static_cast<__bf16>(sacc[r]);
    }
    __builtin_amdgcn_s_waitcnt(FMA_WAIT_LGKM0);
    // phase B: per-row max (lane l reduces row l%32, cols 16*half..)
    {
      float m_part = -1e30f;
#pragma unroll
      for (int i = 0; i < 16; ++i) {
        m_part = fmaxf(m_part, static_cast<float>(
                                   s_s[lane32 * FMA_SROW + 16 * half + i]));
      }
      const float m_tile = fmaxf(m_part, __shfl_xor(m_part, 32));
      const float m_new = fmaxf(m_row, m_tile);
      const float alpha = __expf(m_row - m_new);
      m_row = m_new;
      l_row *= alpha;
      if (half == 0) {
        s_stat[lane32 * 2 + 0] = m_new;
        s_stat[lane32 * 2 + 1] = alpha;
      }
    }
    __builtin_amdgcn_s_waitcnt(FMA_WAIT_LGKM0);
    // phase C: P = exp(S-m) into s_p (MFMA layout) + s_s (for row sums);
    // rescale O by alpha
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r % 4) + 8 * (r / 4) + 4 * half;
      const float m_new = s_stat[row * 2 + 0];
      const float alpha = s_stat[row * 2 + 1];
      const float p = __expf(sacc[r] - m_new);
#pragma unroll
      for (int b = 0; b < kNblk; ++b) oacc[b][r] *= alpha;
      s_p[row * 32 + lane32] = static_cast<__bf16>(p);
    }
    __builtin_amdgcn_s_waitcnt(FMA_WAIT_LGKM0);
    // phase D: row sums of bf16 P straight from s_p (summing the
    // ROUNDED probabilities matches what the PV MFMA actually uses)
    {
      bf16x8_t pr = *reinterpret_cast<const bf16x8_t*>(
          &s_p[lane32 * 32 + 16 * half]);
      bf16x8_t pr2 = *reinterpret_cast<const bf16x8_t*>(
          &s_p[lane32 * 32 + 16 * half + 8]);
      float sum = 0.0f;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        sum += static_cast<float>(pr[i]) + static_cast<float>(pr2[i]);
      }
      l_row += sum + __shfl_xor(sum, 32);
    }
    // O += P . V  (K dim = 32 keys = 2 MFMA k-steps)
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      bf16x8_t pf = *reinterpret_cast<const bf16x8_t*>(
          &s_p[lane32 * 32 + ks2 * 16 + 8 * half]);
#pragma unroll
      for (int b = 0; b < kNblk; ++b) {
        bf16x8_t vf = *reinterpret_cast<const bf16x8_t*>(
            &vdst[(b * 32 + lane32) * 32 + ks2 * 16 + 8 * half]);
        oacc[b] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pf, vf, oacc[b], 0,
                                                          0, 0);
      }
    }
    __builtin_amdgcn_s_waitcnt(FMA_WAIT_LGKM0);  // s_p/s_v reads done
  };

  auto qk = [&](const bf16x8_t(&kf)[HD / 16]) {
    f32x16_t sacc;
#pragma unroll
    for (int r = 0; r < 16; ++r) sacc[r] = 0.0f;
#pragma unroll
    for (int ks = 0; ks < HD / 16; ++ks) {
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf[ks], kf[ks], sacc,
                                                     0, 0, 0);
    }
    return sacc;
  };

  // Software-pipelined key loop: kf is dead the moment the QK MFMAs
  // have consumed it, so the NEXT tile's K loads re-fill the same
  // buffer — and the next V tile streams into the other LDS buffer —
  // while the current tile's softmax+PV runs. The ~700-cycle HBM
  // latency hides behind compute with zero extra registers.
  const int chunk = blockIdx.z;
  const int chunks = gridDim.z;
  if constexpr (NW == 1) {
    // Software-pipelined key loop: kf is dead once the QK MFMAs consumed
    // it, so the NEXT tile's K loads re-fill the same buffer — and the
    // next V tile streams into the other LDS buffer — while the current
    // tile's softmax+PV runs.
    bf16x8_t kf[HD / 16];
    if (chunk < kt_end) {
      prefill_load_k<HD>(K, chunk * 32, t_kv, kv_heads, kvh, lane32, half,
                         kf);
      prefill_stage_v<HD>(V, chunk * 32, t_kv, kv_heads, kvh, lane32, half,
                          s_v[0], c8_lo, c8_hi);
    }
    int par = 0;  // ping-pong parity of the V LDS buffer
    for (int kt = chunk; kt < kt_end; kt += chunks, par ^= 1) {
      f32x16_t sacc = qk(kf);
      if (kt + chunks < kt_end) {
        prefill_load_k<HD>(K, (kt + chunks) * 32, t_kv, kv_heads, kvh,
                           lane32, half, kf);
        prefill_stage_v<HD>(V, (kt + chunks) * 32, t_kv, kv_heads, kvh,
                            lane32, half, s_v[(par ^ 1) % kVBufs], c8_lo,
                            c8_hi);
      }
      softmax_pv(sacc, s_v[par % kVBufs], kt * 32);
    }
  } else if constexpr (PIPE) {
    // Multi-wave pipelined: same K-prefetch + V double buffer as NW==1,
    // with one barrier per tile for the shared V staging.
    bf16x8_t kf[HD / 16];
    if (chunk < kt_end) {
      prefill_load_k<HD>(K, chunk * 32, t_kv, kv_heads, kvh, lane32, half,
                         kf);
      prefill_stage_v<HD>(V, chunk * 32, t_kv, kv_heads, kvh, lane32, half,
                          s_v[0], c8_lo, c8_hi);
    }
    int par = 0;
    for (int kt = chunk; kt < kt_end; kt += chunks, par ^= 1) {
      // (a) all waves' shares of s_v[par] (staged last iteration) are
      // visible; (b) all waves finished reading s_v[par^1] before it is
      // re-staged below
      __syncthreads();
      f32x16_t sacc = qk(kf);
      if (kt + chunks < kt_end) {
        prefill_load_k<HD>(K, (kt + chunks) * 32, t_kv, kv_heads, kvh,
                           lane32, half, kf);
        prefill_stage_v<HD>(V, (kt + chunks) * 32, t_kv, kv_heads, kvh,
                            lane32, half, s_v[(par ^ 1) % kVBufs], c8_lo,
                            c8_hi);
      }
      softmax_pv(sacc, s_v[par % kVBufs], kt * 32);
    }
  } else {
    // Multi-wave streaming: no software prefetch (its live ranges cost
    // ~100 VGPRs), K fragments streamed per 16-k step. One shared V
    // buffer, two barriers per tile.
    for (int kt = chunk; kt < kt_end; kt += chunks) {
      const int key = kt * 32 + lane32;
      const bool klive = key < t_kv;
      const unsigned short* kp =
          K + (static_cast<long long>(klive ? key : 0) * kv_heads + kvh) *
              HD + 8 * half;
      prefill_stage_v<HD>(V, kt * 32, t_kv, kv_heads, kvh, lane32, half,
                          s_v[0], c8_lo, c8_hi);
      f32x16_t sacc;
#pragma unroll
      for (int r = 0; r < 16; ++r) sacc[r] = 0.0f;
#pragma unroll
      for (int ks = 0; ks < HD / 16; ++ks) {
        bf16x8_t kfs;
        if (klive) {
          kfs = *reinterpret_cast<const bf16x8_t*>(kp + ks * 16);
        } else {
#pragma unroll
          for (int i = 0; i < 8; ++i) kfs[i] = static_cast<__bf16>(0.0f);
        }
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf[ks], kfs, sacc,
                                                       0, 0, 0);
      }
      __syncthreads();  // all waves' V shares staged
      softmax_pv(sacc, s_v[0], kt * 32);
      __syncthreads();  // all waves done reading before re-stage
    }
  }

  if (partials != nullptr) {
    // unnormalized partial: [m, l, O(hd)] per row
    const int tiles = gridDim.y;
    float* base = partials +
        ((static_cast<long long>(qh) * tiles + tile) * chunks + chunk) *
            (32 * (HD + 2));
    if (half == 0) {
      base[lane32 * (HD + 2) + 0] = m_row;
      base[lane32 * (HD + 2) + 1] = l_row;
    }
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r % 4) + 8 * (r / 4) + 4 * half;
#pragma unroll
      for (int b = 0; b < kNblk; ++b) {
        base[row * (HD + 2) + 2 + b * 32 + lane32] = oacc[b][r];
      }
    }
    return;
  }

  // ---- epilogue: O /= l, store rows < T
  if (half == 0) s_l[lane32] = l_row;
  __builtin_amdgcn_s_waitcnt(FMA_WAIT_LGKM0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r % 4) + 8 * (r / 4) + 4 * half;
    const int orow = r0 + row;
    if (orow >= T) continue;
    const float lr = s_l[row];
    const float inv_l = lr > 0.0f ? 1.0f / lr : 0.0f;
    unsigned short* op =
        O + (static_cast<long long>(orow) * q_heads + qh) * HD;
#pragma unroll
    for (int b = 0; b < kNblk; ++b) {
      op[b * 32 + lane32] = f32_to_bf16(oacc[b][r] * inv_l);
    }
  }
}

// Merge the per-chunk (m, l, O) partials: log-sum-exp reweighting, then
// normalize and store bf16 (reference for the math: the decode combine
// kernel above — identical recipe, prefill shapes).
__global__ void attn_prefill_combine_kernel(
    const float* __restrict__ partials,  // [qH, tiles, chunks, 32, hd+2]
    unsigned short* __restrict__ O,      // [T, qH, hd]
    int T, int q_heads, int hd, int chunks) {
  const int qh = blockIdx.x;
  const int tile = blockIdx.y;
  const int tiles = gridDim.y;
  const int col = threadIdx.x;  // hd threads
  const float* tbase = partials +
      (static_cast<long long>(qh) * tiles + tile) * chunks *
          (32 * (hd + 2));
  for (int row = 0; row < 32; ++row) {
    const int orow = tile * 32 + row;
    if (orow >= T) continue;
    float m = -1e30f;
    for (int c = 0; c < chunks; ++c) {
      m = fmaxf(m, tbase[(c * 32 + row) * (hd + 2) + 0]);
    }
    float l = 0.0f, acc = 0.0f;
    for (int c = 0; c < chunks; ++c) {
      const float* rb = tbase + (c * 32 + row) * (hd + 2);
      const float w = __expf(rb[0] - m);
      l += rb[1] * w;
      acc += rb[2 + col] * w;
    }
    const float inv_l = l > 0.0f ? 1.0f / l : 0.0f;
    O[(static_cast<long long>(orow) * q_heads + qh) * hd + col] =
        f32_to_bf16(acc * inv_l);
  }
}

}  // namespace

extern "C" int fma_attn_prefill_chunks(int T, int pos0, int q_heads) {
  // Measured on MI355X (tools/prefill_bench.py): splitting the key range
  // never beat the plain path at llama shapes — the partials traffic
  // (qH*tiles*chunks*32*(hd+2) f32) plus the combine launch outweigh the
  // extra occupancy even at T=128 (0.039 ms split vs 0.016 plain). The
  // mechanism stays (correctness-tested) for shapes where it may pay
  // (tiny qH, huge pos0): force via FMA_PREFILL_CHUNKS or the binding.
  (void)T; (void)pos0; (void)q_heads;
  return 1;
}

extern "C" hipError_t fma_launch_attn_prefill_bf16(
    const void* Q, const void* K, const void* V, void* O, int T, int pos0,
    int q_heads, int kv_heads, int hd, float* partials, int chunks,
    hipStream_t stream) {
  if (hd != 64 && hd != 128) return hipErrorInvalidValue;
  if (q_heads % kv_heads != 0) return hipErrorInvalidValue;
  if (chunks > 1 && partials == nullptr) return hipErrorInvalidValue;
  // 4-wave workgroups when 4 consecutive q-heads share one kv-head
  // (llama GQA: group 4/8); else the 1-wave shape
  const int group = q_heads / kv_heads;
  int nw = (q_heads % 4 == 0 && group % 4 == 0) ? 4 : 1;
  if (const char* e = getenv("FMA_PREFILL_NW")) {
    const int forced = atoi(e);
    if (forced == 1 || (forced == 4 && q_heads % 4 == 0 && group % 4 == 0))
      nw = forced;
  }
  dim3 grid(q_heads / nw, (T + 31) / 32, chunks);
  float* par = chunks > 1 ? partials : nullptr;
  const auto* Qp = static_cast<const unsigned short*>(Q);
  const auto* Kp = static_cast<const unsigned short*>(K);
  const auto* Vp = static_cast<const unsigned short*>(V);
  auto* Op = static_cast<unsigned short*>(O);
  const char* pe = getenv("FMA_PREFILL_PIPE");
  const bool pipe = pe != nullptr && atoi(pe) != 0;
  if (hd == 128 && nw == 4) {
    if (pipe) {
      attn_prefill_bf16_kernel<128, 4, true><<<grid, 256, 0, stream>>>(
          Qp, Kp, Vp, Op, par, T, pos0, q_heads, kv_heads);
    } else {
      attn_prefill_bf16_kernel<128, 4><<<grid, 256, 0, stream>>>(
          Qp, Kp, Vp, Op, par, T, pos0, q_heads, kv_heads);
    }
  } else if (hd == 128) {
    attn_prefill_bf16_kernel<128, 1><<<grid, 64, 0, stream>>>(
        Qp, Kp, Vp, Op, par, T, pos0, q_heads, kv_heads);
  } else if (nw == 4) {
    if (pipe) {
      attn_prefill_bf16_kernel<64, 4, true><<<grid, 256, 0, stream>>>(
          Qp, Kp, Vp, Op, par, T, pos0, q_heads, kv_heads);
    } else {
      attn_prefill_bf16_kernel<64, 4><<<grid, 256, 0, stream>>>(
          Qp, Kp, Vp, Op, par, T, pos0, q_heads, kv_heads);
    }
  } else {
    attn_prefill_bf16_kernel<64, 1><<<grid, 64, 0, stream>>>(
        Qp, Kp, Vp, Op, par, T, pos0, q_heads, kv_heads);
  }
  if (chunks > 1) {
    dim3 cgrid(q_heads, (T + 31) / 32);
    attn_prefill_combine_kernel<<<cgrid, hd, 0, stream>>>(
        partials, static_cast<unsigned short*>(O), T, q_heads, hd, chunks);
  }
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// 16-row MFMA prefill: v_mfma_f32_16x16x32_bf16 fragments (lane maps
// hardware-verified by tools/mfma_probe16.hip). Motivation (NOTES.md
// round-3 sketch, landed early): vs the 32-row kernel the accumulator
// footprint halves (oacc kB x 4 f32 = 32 VGPRs at HD=128) and the tile
// grid DOUBLES (T/16 tiles), fixing the mid-T occupancy underfill. The
// C/D layout puts each score row in ONE 16-lane fragment group, so the
// whole online softmax runs on row-local registers + 4 shfl_xor steps —
// no LDS transpose scratch, no cross-half handshakes at all (the 32-row
// kernel burns two LDS round-trips per tile on exactly this).
// Experimental: opt-in via FMA_PREFILL_16=1 (single-chunk path only).

using f32x4_t = __attribute__((ext_vector_type(4))) float;

template <int HD, int NW>
__global__ __launch_bounds__(64 * NW)
__attribute__((amdgpu_waves_per_eu(4)))
void attn_prefill16_bf16_kernel(
    const unsigned short* __restrict__ Q,  // [T, qH, hd]
    const unsigned short* __restrict__ K,  // [S, kvH, hd]
    const unsigned short* __restrict__ V,  // [S, kvH, hd]
    unsigned short* __restrict__ O,        // [T, qH, hd]
    int T, int pos0, int q_heads, int kv_heads) {
  constexpr int kB = HD / 16;  // 16-column output blocks
  const int wid = NW > 1 ? (threadIdx.x >> 6) : 0;
  const int qh = blockIdx.x * NW + wid;
  const int tile = gridDim.y - 1 - blockIdx.y;  // heaviest tiles first
  const int r0 = tile * 16;
  if (r0 >= T) return;  // uniform across the WG
  const int kvh = qh / (q_heads / kv_heads);
  const int l = threadIdx.x & 63;
  const int g = l >> 4;       // 16-lane fragment group, 0..3
  const int c = l & 15;       // position within the group
  const int lane32 = l & 31;  // for the shared V staging helper
  const int half = l >> 5;
  const float scale = rsqrtf(static_cast<float>(HD));

  __shared__ __bf16 s_v[32 * HD];          // V [hd][key], shared by waves
  __shared__ __bf16 s_p_all[NW][16 * 32];  // P tile C->A relayout
  __bf16* const s_p = s_p_all[wid];
  constexpr int kC8PerWave = (HD / 16) / NW;
  const int c8_lo = wid * kC8PerWave, c8_hi = c8_lo + kC8PerWave;

  // Q fragments: lane holds A[c][32*ks + 8*g + i] (A row = q row = c)
  bf16x8_t qf[HD / 32];
  const int q_row = r0 + c;
  const bool row_live = q_row < T;
  {
    const unsigned short* qp =
        Q + (static_cast<long long>(row_live ? q_row : 0) * q_heads + qh) *
                HD + 8 * g;
#pragma unroll
    for (int ks = 0; ks < HD / 32; ++ks) {
      bf16x8_t qv;
      if (row_live) {
        qv = *reinterpret_cast<const bf16x8_t*>(qp + ks * 32);
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) qv[i] = static_cast<__bf16>(0.0f);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        qf[ks][i] = static_cast<__bf16>(static_cast<float>(qv[i]) * scale);
      }
    }
  }

  // Per-row softmax stats in REGISTERS: this lane's reg r owns row
  // 4*g + r, whose 16 key columns live across the lane group — row
  // reductions are 4 shfl_xor steps (masks 1/2/4/8 stay in-group)
  float m_row[4], l_row[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_row[r] = -1e30f;
    l_row[r] = 0.0f;
  }
  f32x4_t oacc[kB];
#pragma unroll
  for (int b = 0; b < kB; ++b) {
#pragma unroll
    for (int r = 0; r < 4; ++r) oacc[b][r] = 0.0f;
  }

  const int t_kv = pos0 + T;  // causal key horizon
  const int kt_end = min((pos0 + r0 + 15) / 32 + 1, (t_kv + 31) / 32);

  for (int kt = 0; kt < kt_end; ++kt) {
    if (NW > 1) __syncthreads();  // all waves done reading s_v
    prefill_stage_v<HD>(V, kt * 32, t_kv, kv_heads, kvh, lane32, half, s_v,
                        c8_lo, c8_hi);
    if (NW > 1) __syncthreads();  // staging visible to every wave
    __builtin_amdgcn_s_waitcnt(FMA_WAIT_LGKM0);

    // S = Q.K^T over the 32-key tile, two 16-key quadrants
    f32x4_t sacc[2];
#pragma unroll
    for (int q2 = 0; q2 < 2; ++q2) {
#pragma unroll
      for (int r = 0; r < 4; ++r) sacc[q2][r] = 0.0f;
      const int key = kt * 32 + 16 * q2 + c;
      const bool klive = key < t_kv;
      const unsigned short* kp =
          K + (static_cast<long long>(klive ? key : 0) * kv_heads + kvh) *
                  HD + 8 * g;
#pragma unroll
      for (int ks = 0; ks < HD / 32; ++ks) {
        bf16x8_t kf;
        if (klive) {
          kf = *reinterpret_cast<const bf16x8_t*>(kp + ks * 32);
        } else {
#pragma unroll
          for (int i = 0; i < 8; ++i) kf[i] = static_cast<__bf16>(0.0f);
        }
        sacc[q2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[ks], kf,
                                                           sacc[q2], 0, 0, 0);
      }
    }

    // Online softmax, all register-resident
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = 4 * g + r;
      const int qpos = pos0 + r0 + row;
      const int k0 = kt * 32 + c, k1 = kt * 32 + 16 + c;
      if (k0 > qpos || k0 >= t_kv) sacc[0][r] = -1e30f;
      if (k1 > qpos || k1 >= t_kv) sacc[1][r] = -1e30f;
      float m = fmaxf(sacc[0][r], sacc[1][r]);
      m = fmaxf(m, __shfl_xor(m, 1));
      m = fmaxf(m, __shfl_xor(m, 2));
      m = fmaxf(m, __shfl_xor(m, 4));
      m = fmaxf(m, __shfl_xor(m, 8));
      const float m_new = fmaxf(m_row[r], m);
      const float alpha = __expf(m_row[r] - m_new);
      m_row[r] = m_new;
      l_row[r] *= alpha;
#pragma unroll
      for (int b = 0; b < kB; ++b) oacc[b][r] *= alpha;
      const __bf16 p0 = static_cast<__bf16>(__expf(sacc[0][r] - m_new));
      const __bf16 p1 = static_cast<__bf16>(__expf(sacc[1][r] - m_new));
      s_p[row * 32 + c] = p0;
      s_p[row * 32 + 16 + c] = p1;
      // sum the bf16-ROUNDED probabilities: matches what the PV MFMA uses
      float sum = static_cast<float>(p0) + static_cast<float>(p1);
      sum += __shfl_xor(sum, 1);
      sum += __shfl_xor(sum, 2);
      sum += __shfl_xor(sum, 4);
      sum += __shfl_xor(sum, 8);
      l_row[r] += sum;
    }
    __builtin_amdgcn_s_waitcnt(FMA_WAIT_LGKM0);  // s_p relayout (wave-local)

    // O += P.V: K-dim 32 = exactly ONE MFMA step per 16-column block
    bf16x8_t pf = *reinterpret_cast<const bf16x8_t*>(&s_p[c * 32 + 8 * g]);
#pragma unroll
    for (int b = 0; b < kB; ++b) {
      bf16x8_t vf = *reinterpret_cast<const bf16x8_t*>(
          &s_v[(b * 16 + c) * 32 + 8 * g]);
      oacc[b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, oacc[b], 0,
                                                        0, 0);
    }
    __builtin_amdgcn_s_waitcnt(FMA_WAIT_LGKM0);  // s_p/s_v reads done
  }

  // Epilogue: l_row is already uniform across the lane group — no LDS
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = r0 + 4 * g + r;
    if (orow >= T) continue;
    const float inv = l_row[r] > 0.0f ? 1.0f / l_row[r] : 0.0f;
    unsigned short* op =
        O + (static_cast<long long>(orow) * q_heads + qh) * HD;
#pragma unroll
    for (int b = 0; b < kB; ++b) {
      op[b * 16 + c] = f32_to_bf16(oacc[b][r] * inv);
    }
  }
}

extern "C" hipError_t fma_launch_attn_prefill16_bf16(
    const void* Q, const void* K, const void* V, void* O, int T, int pos0,
    int q_heads, int kv_heads, int hd, hipStream_t stream) {
  if (hd != 64 && hd != 128) return hipErrorInvalidValue;
  if (q_heads % kv_heads != 0) return hipErrorInvalidValue;
  const int group = q_heads / kv_heads;
  int nw = (q_heads % 4 == 0 && group % 4 == 0) ? 4 : 1;
  if (const char* e = getenv("FMA_PREFILL_NW")) {
    const int forced = atoi(e);
    if (forced == 1 || (forced == 4 && q_heads % 4 == 0 && group % 4 == 0))
      nw = forced;
  }
  dim3 grid(q_heads / nw, (T + 15) / 16);
  const auto* Qp = static_cast<const unsigned short*>(Q);
  const auto* Kp = static_cast<const unsigned short*>(K);
  const auto* Vp = static_cast<const unsigned short*>(V);
  auto* Op = static_cast<unsigned short*>(O);
  if (hd == 128 && nw == 4) {
    attn_prefill16_bf16_kernel<128, 4><<<grid, 256, 0, stream>>>(
        Qp, Kp, Vp, Op, T, pos0, q_heads, kv_heads);
  } else if (hd == 128) {
    attn_prefill16_bf16_kernel<128, 1><<<grid, 64, 0, stream>>>(
        Qp, Kp, Vp, Op, T, pos0, q_heads, kv_heads);
  } else if (nw == 4) {
    attn_prefill16_bf16_kernel<64, 4><<<grid, 256, 0, stream>>>(
        Qp, Kp, Vp, Op, T, pos0, q_heads, kv_heads);
  } else {
    attn_prefill16_bf16_kernel<64, 1><<<grid, 64, 0, stream>>>(
        Qp, Kp, Vp, Op, T, pos0, q_heads, kv_heads);
  }
  return hipGetLastError();
}
