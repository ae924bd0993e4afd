// Shared helpers for the chainermn_amd gfx950 kernels.
//
// Design rules (see /opt/skills/guides/cdna_hip_programming.md):
//  * wave64; block size 256 (4 waves)
//  * memory-bound kernels: grid-stride, capped grid (G11), 16 B/lane
//    vectorized access (G13) — bf16/f16 moved as uint4 (8 elems), fp32 as
//    two float4 (8 elems)
//  * fp32 accumulation everywhere
#pragma once

#include <hip/hip_runtime.h>
#include <c10/util/BFloat16.h>
#include <c10/util/Half.h>

#define CMN_BLOCK 256
#define CMN_WAVE 64

// Dispatch over the dtypes these kernels support (no double).
#define CMN_DISPATCH(TYPE, NAME, ...)                                        \
  [&] {                                                                      \
    switch (TYPE) {                                                          \
      case at::ScalarType::Float: {                                          \
        using scalar_t = float;                                              \
        return __VA_ARGS__();                                                \
      }                                                                      \
      case at::ScalarType::BFloat16: {                                       \
        using scalar_t = c10::BFloat16;                                      \
        return __VA_ARGS__();                                                \
      }                                                                      \
      case at::ScalarType::Half: {                                           \
        using scalar_t = c10::Half;                                          \
        return __VA_ARGS__();                                                \
      }                                                                      \
      default:                                                               \
        TORCH_CHECK(false, NAME, ": unsupported dtype");                     \
    }                                                                        \
  }()

static inline int cmn_grid(long items, int per_block) {
  long blocks = (items + per_block - 1) / per_block;
  if (blocks < 1) blocks = 1;
  if (blocks > 2048) blocks = 2048;  // grid-stride the rest
  return (int)blocks;
}

// ---- 8-element vector load/store: float[8] <-> native dtype -------------
template <typename T>
struct V8;

template <>
struct V8<float> {
  static __device__ __forceinline__ void load(const float* p, float (&r)[8]) {
    const float4 a = *reinterpret_cast<const float4*>(p);
    const float4 b = *reinterpret_cast<const float4*>(p + 4);
    r[0] = a.x; r[1] = a.y; r[2] = a.z; r[3] = a.w;
    r[4] = b.x; r[5] = b.y; r[6] = b.z; r[7] = b.w;
  }
  static __device__ __forceinline__ void store(float* p, const float (&r)[8]) {
    *reinterpret_cast<float4*>(p) = make_float4(r[0], r[1], r[2], r[3]);
    *reinterpret_cast<float4*>(p + 4) = make_float4(r[4], r[5], r[6], r[7]);
  }
};

template <typename T16>
struct V8_16bit {
  static __device__ __forceinline__ void load(const T16* p, float (&r)[8]) {
    const uint4 v = *reinterpret_cast<const uint4*>(p);
    const unsigned u[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      T16 lo, hi;
      lo = T16(uint16_t(u[k] & 0xffffu), typename T16::from_bits_t());
      hi = T16(uint16_t(u[k] >> 16), typename T16::from_bits_t());
      r[2 * k] = float(lo);
      r[2 * k + 1] = float(hi);
    }
  }
  static __device__ __forceinline__ void store(T16* p, const float (&r)[8]) {
    uint4 v;
    unsigned u[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      const uint16_t lo = T16(r[2 * k]).x;
      const uint16_t hi = T16(r[2 * k + 1]).x;
      u[k] = unsigned(lo) | (unsigned(hi) << 16);
    }
    v.x = u[0]; v.y = u[1]; v.z = u[2]; v.w = u[3];
    *reinterpret_cast<uint4*>(p) = v;
  }
};

template <>
struct V8<c10::BFloat16> : V8_16bit<c10::BFloat16> {};
template <>
struct V8<c10::Half> : V8_16bit<c10::Half> {};

// ---- wave + block reduction (sum) ---------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_down(v, off, CMN_WAVE);
  return v;
}

// Block-level sum; `scratch` must hold >= BLOCK/WAVE floats; result valid
// on thread 0.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (CMN_WAVE - 1);
  const int wid = threadIdx.x / CMN_WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = 0.f;
  if (wid == 0) {
    const int nw = blockDim.x / CMN_WAVE;
    r = (lane < nw) ? scratch[lane] : 0.f;
    r = wave_reduce_sum(r);
  }
  __syncthreads();
  return r;
}

// Binary search: greatest i with offsets[i] <= idx; offsets ascending.
__device__ __forceinline__ int find_segment(const long* offsets, int n,
                                            long idx) {
  int lo = 0, hi = n - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (offsets[mid] <= idx) lo = mid; else hi = mid - 1;
  }
  return lo;
}
