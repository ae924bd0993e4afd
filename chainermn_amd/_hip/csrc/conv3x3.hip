// EXPERIMENTAL: 3x3 / stride-1 / pad-1 forward convolution as implicit
// GEMM on MFMA (gfx950), channels_last.
//
// Motivation (round-2 groundwork): MIOpen's asm igemm kernels for these
// shapes use the CDNA2-era 32x32x8 MFMA — half the 32x32x16/16x16x32 rate
// on gfx950.  This kernel reuses gemm.hip's 128x128x64 tile but builds
// the A-operand (im2col view of the input) on the fly:
//
//   y[m, co] = sum_{ky,kx,ci} x[n, oh+ky-1, ow+kx-1, ci] * w[co,ky,kx,ci]
//   M = N*H*W (m -> n,oh,ow),  K = 9*Cin (k -> (ky,kx), ci)
//
// Cin % 64 == 0 keeps every 64-wide K-tile inside one (ky,kx) tap.  The
// input tile is staged through registers with per-lane boundary
// predication (padding rows/cols load zeros) and written to LDS at the
// XOR-swizzled address directly (register staging may scatter, unlike
// global_load_lds); the weight tile uses glds with a pre-swizzled source
// like gemm.hip.  Single-buffered: measured 280-425 TF vs MIOpen's ~460;
// a naive 2-buffer in-loop staging variant measured SLOWER (ordinary
// register loads beside in-flight glds hit hipcc's vmcnt(0) drain — the
// programming guide's mixed-load-kinds trap), so the pipelined version
// needs the T14 split (issue loads early, ds_write after the barrier) —
// round-2 work, see STATUS.md.
//
// Not wired into any model path; exposed as ops-level experimental API.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define C3_BM 128
#define C3_BN 128
#define C3_BK 64
#define C3_THREADS 256

__device__ __forceinline__ int c3_swz(int row, int within_byte) {
  return within_byte ^ ((row & 7) << 4);
}

__global__ __launch_bounds__(C3_THREADS) void conv3x3s1_fwd_kernel(
    const short* __restrict__ X,  // [N, H, W, Cin] bf16 bits
    const short* __restrict__ Wt, // [Cout, 3, 3, Cin]
    short* __restrict__ Y,        // [N, H, W, Cout]
    int Nb, int H, int Wd, int Cin, int Cout) {
  __shared__ __attribute__((aligned(16))) char lds[2 * C3_BM * C3_BK * 2];
  char* ldsA = lds;
  char* ldsB = lds + C3_BM * C3_BK * 2;

  const long M = (long)Nb * H * Wd;
  const int ntile_n = Cout / C3_BN;
  const int tile_m = blockIdx.x / ntile_n;
  const int tile_n = blockIdx.x % ntile_n;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;
  const int KW = 9 * Cin;

  f32x4 acc[4][4];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int nkt = KW / C3_BK;
  for (int kt = 0; kt < nkt; ++kt) {
    const int k0 = kt * C3_BK;
    const int r = k0 / Cin;          // tap index 0..8
    const int ky = r / 3, kx = r % 3;
    const int ci0 = k0 - r * Cin;
    // ---- stage A: register loads with boundary predication, swizzled
    // ds_write (2 chunks of 16 B per thread cover 128x64 bf16) ----
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int p = (i * C3_THREADS + t) * 16;
      const int row = p >> 7;
      const int within = p & 127;
      const long m = (long)tile_m * C3_BM + row;
      uint4 v = {0u, 0u, 0u, 0u};
      if (m < M) {
        const long n = m / ((long)H * Wd);
        const long rem = m - n * (long)H * Wd;
        const int oh = (int)(rem / Wd);
        const int ow = (int)(rem - (long)oh * Wd);
        const int ih = oh + ky - 1;
        const int iw = ow + kx - 1;
        if (ih >= 0 && ih < H && iw >= 0 && iw < Wd) {
          const short* src = X + (((n * H + ih) * (long)Wd + iw) * Cin
                                  + ci0 + (within >> 1));
          v = *reinterpret_cast<const uint4*>(src);
        }
      }
      *reinterpret_cast<uint4*>(ldsA + row * 128 + c3_swz(row, within)) = v;
    }
    // ---- stage B: weights [Cout][9][Cin]; row = co within tile ----
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      // glds needs a WAVE-UNIFORM LDS base (+ lane*16 implicit)
      const int base = (i * C3_THREADS + wave * 64) * 16;
      const int p = base + lane * 16;
      const int row = p >> 7;
      const int within = p & 127;
      const short* src = Wt + ((long)(tile_n * C3_BN + row)) * KW + k0
                         + (c3_swz(row, within) >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              (ldsB + base),
          16, 0, 0);
    }
    __syncthreads();
    const int fr = lane & 15;
    const int kq = lane >> 4;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const int row = wr * 64 + m * 16 + fr;
        const int kbyte = ks * 64 + kq * 16;
        a_frag[m] = *(const bf16x8*)(ldsA + row * 128 + c3_swz(row, kbyte));
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int row = wc * 64 + n * 16 + fr;
        const int kbyte = ks * 64 + kq * 16;
        b_frag[n] = *(const bf16x8*)(ldsB + row * 128 + c3_swz(row, kbyte));
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    }
    __syncthreads();
  }

  const int col0 = tile_n * C3_BN + wc * 64 + (lane & 15);
  const long row_base = (long)tile_m * C3_BM + wr * 64 + (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long row = row_base + m * 16 + rg;
      if (row >= M) continue;
#pragma unroll
      for (int n = 0; n < 4; ++n)
        Y[row * Cout + col0 + n * 16] =
            (short)c10::BFloat16(acc[m][n][rg]).x;
    }
  }
}

}  // namespace

torch::Tensor conv3x3s1_fwd_bf16(torch::Tensor x, torch::Tensor w) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16
              && w.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.dim() == 4
              && x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "x must be channels_last NCHW");
  TORCH_CHECK(w.dim() == 4 && w.size(2) == 3 && w.size(3) == 3
              && w.is_contiguous(at::MemoryFormat::ChannelsLast),
              "w must be channels_last [Cout, Cin, 3, 3]");
  const long Nb = x.size(0);
  const int Cin = (int)x.size(1), H = (int)x.size(2), Wd = (int)x.size(3);
  const int Cout = (int)w.size(0);
  TORCH_CHECK(Cin % 64 == 0 && Cout % C3_BN == 0,
              "requires Cin%64==0 and Cout%128==0");
  const long M = Nb * H * Wd;
  TORCH_CHECK(M % C3_BM == 0, "requires N*H*W % 128 == 0");
  auto y = torch::empty({Nb, Cout, H, Wd},
                        x.options().memory_format(
                            at::MemoryFormat::ChannelsLast));
  auto stream = c10::hip::getCurrentHIPStream();
  const long grid = (M / C3_BM) * (Cout / C3_BN);
  hipLaunchKernelGGL(conv3x3s1_fwd_kernel, dim3((unsigned)grid),
                     dim3(C3_THREADS), 0, stream.stream(),
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     (short*)y.data_ptr(), (int)Nb, H, Wd, Cin, Cout);
  return y;
}
