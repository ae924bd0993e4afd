// Hand-written MFMA bf16 GEMM (gfx950): D[M,N] = A[M,K] @ B[N,K]^T.
//
// The MFMA/LDS-tiled building block for GEMM-shaped fused ops (nn.Linear
// runs y = x @ W^T with W stored [N,K], so the B^T form is the native
// layout).  Structure per /opt/skills/guides/cdna_hip_programming.md §5:
//
//  * 128x128 block tile, BK=64; 4 waves as 2x2, each owning a 64x64
//    sub-tile as 4x4 fragments of v_mfma_f32_16x16x32_bf16 (per-wave MFMA,
//    accumulators in the unified VGPR/AGPR file)
//  * global -> LDS staging via __builtin_amdgcn_global_load_lds width 16
//    (direct-to-LDS DMA, lane-linear destination), double-buffered
//  * LDS bank-conflict fix: XOR swizzle byte ^= ((row&7)<<4) applied to
//    the per-lane *source* address (glds writes lane-linear, rule: linear
//    dest + inverse-swizzled source + swizzle on read) and to every
//    ds_read_b128 offset
//  * XCD-aware bijective blockIdx swizzle (8 XCDs with private L2s)
//
// Correctness-first: requires M%128 == N%128 == K%64 == 0 (the Python
// wrapper falls back to hipBLASLt otherwise).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define BM 128
#define BN 128
#define BK 64
#define THREADS 256

__device__ __forceinline__ int swz(int row, int within_byte) {
  return within_byte ^ ((row & 7) << 4);
}

__global__ __launch_bounds__(THREADS) void gemm_bt_bf16_kernel(
    const short* __restrict__ A,  // [M, K] bf16 bits
    const short* __restrict__ B,  // [N, K] bf16 bits
    short* __restrict__ D,        // [M, N] bf16 bits
    int M, int N, int K) {
  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM * BK * 2];
  // layout: [buf][mat][row][swizzled 128-byte row]
  auto ldsA = [&](int buf) -> char* {
    return lds + buf * 2 * BM * BK * 2;
  };
  auto ldsB = [&](int buf) -> char* {
    return lds + buf * 2 * BM * BK * 2 + BM * BK * 2;
  };

  // XCD-aware bijective remap (guide §5: nwg may not divide 8)
  const int nwg = gridDim.x;
  const int orig = blockIdx.x;
  const int q = nwg / 8, r = nwg % 8, xcd = orig % 8;
  const int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                   + orig / 8;
  const int ntile_n = N / BN;
  const int tile_m = wgid / ntile_n;
  const int tile_n = wgid % ntile_n;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1;   // wave row (0..1)
  const int wc = wave & 1;    // wave col (0..1)

  // ---- staging: 4 glds per wave per matrix; wave-uniform LDS base,
  // per-lane swizzled global source ----
  auto stage = [&](int buf, int kt) {
    const long k0 = (long)kt * BK;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      // A: wave handles LDS bytes [(i*4+wave)*1024, +1024)
      const int p = (i * 4 + wave) * 1024 + lane * 16;
      const int row = p >> 7;
      const int within = p & 127;
      const short* srcA = A + ((long)(tile_m * BM + row)) * K + k0
                          + (swz(row, within) >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)
              (ldsA(buf) + (i * 4 + wave) * 1024),
          16, 0, 0);
      const short* srcB = B + ((long)(tile_n * BN + row)) * K + k0
                          + (swz(row, within) >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcB,
          (__attribute__((address_space(3))) unsigned int*)
              (ldsB(buf) + (i * 4 + wave) * 1024),
          16, 0, 0);
    }
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int nkt = K / BK;
  stage(0, 0);
  __syncthreads();  // waits the in-flight LDS-DMA (vmcnt0) + barrier

  for (int kt = 0; kt < nkt; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < nkt) stage(buf ^ 1, kt + 1);
    // fragment row/col for this lane
    const int fr = lane & 15;          // row within 16
    const int kq = lane >> 4;          // k-quarter (0..3): 8 elems each
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {   // two K=32 steps per K-tile
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const int row = wr * 64 + m * 16 + fr;
        const int kbyte = ks * 64 + kq * 16;
        a_frag[m] = *(const bf16x8*)(ldsA(buf) + row * 128
                                     + swz(row, kbyte));
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int row = wc * 64 + n * 16 + fr;
        const int kbyte = ks * 64 + kq * 16;
        b_frag[n] = *(const bf16x8*)(ldsB(buf) + row * 128
                                     + swz(row, kbyte));
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    }
    __syncthreads();  // drains next-tile DMA (vmcnt0) + joins waves
  }

  // ---- epilogue: C/D map for 16x16x32: col = lane&15,
  // row = (lane>>4)*4 + reg ----
  const int col0 = tile_n * BN + wc * 64 + (lane & 15);
  const int row_base = tile_m * BM + wr * 64 + (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long row = row_base + m * 16 + rg;
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const float v = acc[m][n][rg];
        D[row * N + col0 + n * 16] =
            (short)c10::BFloat16(v).x;
      }
    }
  }
}

}  // namespace

torch::Tensor gemm_bt_bf16(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16
              && b.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1));
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK == 0,
              "gemm_bt_bf16 requires M%128==N%128==K%64==0");
  auto d = torch::empty({M, N}, a.options());
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = (M / BM) * (N / BN);
  hipLaunchKernelGGL(gemm_bt_bf16_kernel, dim3(grid), dim3(THREADS), 0,
                     stream.stream(), (const short*)a.data_ptr(),
                     (const short*)b.data_ptr(), (short*)d.data_ptr(), M,
                     N, K);
  return d;
}
