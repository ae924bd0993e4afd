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

#define CMN_STATS_BUCKETS 16

// BN-stats epilogue (shared by conv3x3 + gemm256): per-channel
// (sum y, sum y^2) of the block's bf16-rounded output, reduced
// wave -> block (LDS) and atomically added into SP[bucket][2*Cout]
// (CMN_STATS_BUCKETS buckets hashed by blockIdx bound the per-address
// atomic chains; the Python side sums the 16 rows).  Saves the
// standalone bn_reduce read pass over the activation for the BN that
// consumes this output.  Requires a barrier before (LDS reuse) — the
// caller provides it.  (Defined here so conv3x3.hip and gemm256.hip share it.)
template <int MF, int WAVES_M, int WAVES_N, typename ACC>
__device__ __forceinline__ void stats_epilogue(
    float* __restrict__ SP, char* lds, const ACC& acc, int lane, int wave,
    int wr, int wc, int tile_n, int Cout, unsigned bid) {
  constexpr int BN = 64 * WAVES_N;
  float s[4], qq[4];
#pragma unroll
  for (int n = 0; n < 4; ++n) { s[n] = 0.f; qq[n] = 0.f; }
#pragma unroll
  for (int m = 0; m < 2 * MF; ++m)
#pragma unroll
    for (int rg = 0; rg < 4; ++rg)
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const float v = float(c10::BFloat16(acc[m][n][rg]));
        s[n] += v;
        qq[n] += v * v;
      }
#pragma unroll
  for (int n = 0; n < 4; ++n) {
    s[n] += __shfl_xor(s[n], 16);
    s[n] += __shfl_xor(s[n], 32);
    qq[n] += __shfl_xor(qq[n], 16);
    qq[n] += __shfl_xor(qq[n], 32);
  }
  float* sl = reinterpret_cast<float*>(lds);  // [wave][2][64]
  if (lane < 16) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      sl[wave * 128 + lane + n * 16] = s[n];
      sl[wave * 128 + 64 + lane + n * 16] = qq[n];
    }
  }
  __syncthreads();
  if (wr == 0) {  // one wave per distinct 64-col set; one lane per col
    float ss = 0.f, qs = 0.f;
#pragma unroll
    for (int k = 0; k < WAVES_M; ++k) {
      const int w2 = k * WAVES_N + wc;
      ss += sl[w2 * 128 + lane];
      qs += sl[w2 * 128 + 64 + lane];
    }
    const int c = tile_n * BN + wc * 64 + lane;
    const long base = (long)(bid & (CMN_STATS_BUCKETS - 1)) * 2 * Cout;
    atomicAdd(&SP[base + c], ss);
    atomicAdd(&SP[base + Cout + c], qs);
  }
}


// BN-BACKWARD reduce fused into a dgrad epilogue: the kernel that
// PRODUCES gy for a BN(+ReLU) layer's backward also reduces
// (sum g' | sum g'*xhat) with g' = gy * relu_mask(xhat*gamma+beta),
// xhat = (x - mean) * rstd — the standalone bn_bwd_reduce read pass
// over (x, gy) collapses to one x read here (gy is in registers).
// Same wave->block->bucketed-atomics shape as stats_epilogue.
struct BnBwdArgs {
  const short* x;        // BN input, channels_last [M, C] bf16 bits
  const float* mean;
  const float* rstd;
  const short* gamma;    // bf16 bits
  const short* beta;
  float* partials;       // [CMN_STATS_BUCKETS][2*C], pre-zeroed
  const short* d;        // the produced gy tile base (set by the kernel)
};

template <int MF, int WAVES_M, int WAVES_N, typename ACC>
__device__ __forceinline__ void bnbwd_epilogue(
    const BnBwdArgs& bn, char* lds, const ACC& acc, int lane, int wave,
    int wr, int wc, long row_base, int tile_n, int Cout, unsigned bid) {
  // COALESCED tile sweep: instead of walking the accumulator fragments
  // (whose lane layout forces per-element x reads), re-read the block's
  // just-written D tile (L2-hot) alongside x with 16-byte loads and
  // accumulate per-channel (sum g' | sum g'*xhat) via LDS atomics.
  // acc/wr/wc unused in this variant; row_base/tile_n locate the tile.
  (void)acc; (void)wr; (void)wc;
  constexpr int BN = 64 * WAVES_N;
  const int t = wave * 64 + lane;
  // thread -> fixed channel octet; (512 / (BN/8)) threads share one
  const int g = t % (BN / 8);
  const int tr = t / (BN / 8);
  constexpr int RPT = 256 / (512 / (BN / 8));  // rows per thread
  const long row0 = row_base;          // tile_m * 256 (caller passes)
  const int c0 = tile_n * BN + g * 8;
  float mn[8], rs[8], gm[8], bt[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    mn[k] = bn.mean[c0 + k];
    rs[k] = bn.rstd[c0 + k];
    gm[k] = float(reinterpret_cast<const c10::BFloat16*>(bn.gamma)[c0 + k]);
    bt[k] = float(reinterpret_cast<const c10::BFloat16*>(bn.beta)[c0 + k]);
  }
  float s[8], qq[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) { s[k] = 0.f; qq[k] = 0.f; }
  for (int r = 0; r < RPT; ++r) {
    const long row = row0 + tr * RPT + r;
    const long off = row * Cout + c0;
    const uint4 xv4 = *reinterpret_cast<const uint4*>(
        reinterpret_cast<const short*>(bn.x) + off);
    const uint4 dv4 = *reinterpret_cast<const uint4*>(bn.d + off);
    const unsigned* xu = reinterpret_cast<const unsigned*>(&xv4);
    const unsigned* du = reinterpret_cast<const unsigned*>(&dv4);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      const unsigned xbits = (k & 1) ? (xu[k >> 1] >> 16)
                                     : (xu[k >> 1] & 0xFFFFu);
      const unsigned dbits = (k & 1) ? (du[k >> 1] >> 16)
                                     : (du[k >> 1] & 0xFFFFu);
      c10::BFloat16 xb, db;
      xb.x = (unsigned short)xbits;
      db.x = (unsigned short)dbits;
      const float xh = (float(xb) - mn[k]) * rs[k];
      const bool mask = (xh * gm[k] + bt[k]) > 0.f;
      const float gv = mask ? float(db) : 0.f;
      s[k] += gv;
      qq[k] += gv * xh;
    }
  }
  // per-channel accumulation across the threads sharing an octet
  float* sl = reinterpret_cast<float*>(lds);  // [2 * BN]
  for (int i = t; i < 2 * BN; i += 512) sl[i] = 0.f;
  __syncthreads();
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    atomicAdd(&sl[g * 8 + k], s[k]);
    atomicAdd(&sl[BN + g * 8 + k], qq[k]);
  }
  __syncthreads();
  if (t < BN) {
    const long base = (long)(bid & (CMN_STATS_BUCKETS - 1)) * 2 * Cout;
    atomicAdd(&bn.partials[base + tile_n * BN + t], sl[t]);
    atomicAdd(&bn.partials[base + Cout + tile_n * BN + t], sl[BN + t]);
  }
}
