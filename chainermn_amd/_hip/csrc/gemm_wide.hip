// Narrow-N variants of the phased 256-row MFMA bf16 GEMM:
// D[M,N] = A[M,K] @ B[N,K]^T for N % 128 == 0 (BN=128) or N % 64 == 0
// (BN=64) — the ResNet 1x1-conv shapes whose output-channel count is
// too small for the 256x256 tile (gemm256.hip) and which previously
// fell back to the 128x128 2-barrier kernel (~440 TF on the tall-skinny
// shapes).  Same phased schedule and wave geometries as conv3x3.hip
// (which measured 738 TF at the BN=128 geometry with im2col on top):
//
//   BN=128: waves 4(M)x2(N), per-wave  64x64
//   BN= 64: waves 8x1,       per-wave  32x64
//
// Optional fused BN-stats partials (stats_epilogue, common.h) so the
// 1x1-conv -> BN path keeps its producer-fused statistics on these
// shapes too.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define GW_THREADS 512

__device__ __forceinline__ int gw_swz(int row, int within_byte) {
  return within_byte ^ ((row & 7) << 4);
}

template <int N>
__device__ __forceinline__ void gw_waitbar() {
  if constexpr (N == 0)
    asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
  else if constexpr (N == 1)
    asm volatile("s_waitcnt vmcnt(1)\n\ts_barrier" ::: "memory");
  else if constexpr (N == 2)
    asm volatile("s_waitcnt vmcnt(2)\n\ts_barrier" ::: "memory");
  else if constexpr (N == 3)
    asm volatile("s_waitcnt vmcnt(3)\n\ts_barrier" ::: "memory");
  else if constexpr (N == 4)
    asm volatile("s_waitcnt vmcnt(4)\n\ts_barrier" ::: "memory");
  else
    asm volatile("s_barrier" ::: "memory");
}

template <int WM, int WAVES_M, int WAVES_N, bool ACC = false>
__global__ __launch_bounds__(GW_THREADS, 1) void gemm_btw_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ D, float* __restrict__ SP, BnBwdArgs BNA,
    int M, int N, int K) {
  constexpr int BM = 256;
  constexpr int BN = 64 * WAVES_N;
  constexpr int MF = WM / 32;
  constexpr int SA = 2;
  constexpr int SB = (WAVES_N >= 2) ? WAVES_N / 2 : 1;
  constexpr bool B_BOTH = (WAVES_N == 1);
  constexpr int W_P0 = B_BOTH ? SA : (SB + SA);
  constexpr int W_P1 = B_BOTH ? -1 : (SA + SB);
  constexpr int W_P3 = B_BOTH ? SA : (SA + SB);
  constexpr int P_P0 = B_BOTH ? 0 : SB;
  constexpr int P_P1 = B_BOTH ? -1 : 0;

  __shared__ __attribute__((aligned(16))) char lds[2 * (BM + BN) * 64 * 2];
  auto ldsA = [&](int buf) -> char* {
    return lds + buf * (BM + BN) * 64 * 2;
  };
  auto ldsB = [&](int buf) -> char* {
    return lds + buf * (BM + BN) * 64 * 2 + BM * 64 * 2;
  };

  const int nwg = gridDim.x;
  const int orig = blockIdx.x;
  const int q = nwg / 8, r = nwg % 8, xcd = orig % 8;
  const int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                   + orig / 8;
  const int ntile_n = N / BN;
  const long tile_m = wgid / ntile_n;
  const int tile_n = wgid % ntile_n;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave / WAVES_N;
  const int wc = wave % WAVES_N;
  const int fr = lane & 15;
  const int kq = lane >> 4;

  // A-half h = the h-th (WM/2)-row half of every wave-row block
  auto stage_a = [&](int buf, int kt, int h) {
    const long k0 = (long)kt * 64;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int c = i * 8 + wave;
      constexpr int CPW = WM / 16;
      const int row0 = (c / CPW) * WM + h * (WM / 2) + (c % CPW) * 8;
      const int p = lane * 16;
      const int row = row0 + (p >> 7);
      const int within = p & 127;
      const short* src = A + (tile_m * BM + row) * (long)K + k0
                         + (gw_swz(row, within) >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              (ldsA(buf) + row0 * 128),
          16, 0, 0);
    }
  };

  auto stage_b = [&](int buf, int kt, int h) {
    const long k0 = (long)kt * 64;
#pragma unroll
    for (int i = 0; i < SB; ++i) {
      int row0;
      if constexpr (B_BOTH) {
        row0 = wave * 8;
      } else {
        const int c = i * 8 + wave;
        row0 = (c / 4) * 64 + h * 32 + (c & 3) * 8;
      }
      const int p = lane * 16;
      const int row = row0 + (p >> 7);
      const int within = p & 127;
      const short* src = B + ((long)(tile_n * BN + row)) * K + k0
                         + (gw_swz(row, within) >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              (ldsB(buf) + row0 * 128),
          16, 0, 0);
    }
  };

  f32x4 acc[2 * MF][4];
#pragma unroll
  for (int m = 0; m < 2 * MF; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  bf16x8 af[2][MF];
  bf16x8 bfr[2][2];

#define GW_READ_A(buf, qa)                                                   \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int mi = 0; mi < MF; ++mi) {                    \
    const int row = wr * WM + (qa) * (WM / 2) + mi * 16 + fr;                \
    af[ks][mi] = *(const bf16x8*)(ldsA(buf) + row * 128                      \
                                  + gw_swz(row, ks * 64 + kq * 16));         \
  }
#define GW_READ_B(buf, qb)                                                   \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int ni = 0; ni < 2; ++ni) {                     \
    const int row = wc * 64 + (qb) * 32 + ni * 16 + fr;                      \
    bfr[ks][ni] = *(const bf16x8*)(ldsB(buf) + row * 128                     \
                                   + gw_swz(row, ks * 64 + kq * 16));        \
  }
#define GW_MFMA(qa, qb)                                                      \
  __builtin_amdgcn_s_setprio(1);                                            \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int mi = 0; mi < MF; ++mi)                      \
          _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                   \
    acc[(qa) * MF + mi][(qb) * 2 + ni] =                                     \
        __builtin_amdgcn_mfma_f32_16x16x32_bf16(                             \
            af[ks][mi], bfr[ks][ni], acc[(qa) * MF + mi][(qb) * 2 + ni],     \
            0, 0, 0);                                                        \
  __builtin_amdgcn_s_setprio(0);
#define GW_BAR() asm volatile("s_barrier" ::: "memory");

  const int nkt = K / 64;
  stage_a(0, 0, 0);
  stage_b(0, 0, 0);
  stage_a(0, 0, 1);
  if constexpr (!B_BOTH) stage_b(0, 0, 1);
  gw_waitbar<B_BOTH ? SA : SA + SB>();

  for (int kt = 0; kt < nkt - 1; ++kt) {
    const int buf = kt & 1;
    const int nxt = buf ^ 1;
    stage_a(nxt, kt + 1, 0);
    GW_READ_A(buf, 0)
    GW_READ_B(buf, 0)
    gw_waitbar<W_P0>();
    GW_MFMA(0, 0)
    stage_b(nxt, kt + 1, 0);
    GW_READ_A(buf, 1)
    gw_waitbar<W_P1>();
    GW_MFMA(1, 0)
    stage_a(nxt, kt + 1, 1);
    GW_READ_B(buf, 1)
    GW_BAR()
    GW_MFMA(1, 1)
    if constexpr (!B_BOTH) stage_b(nxt, kt + 1, 1);
    GW_READ_A(buf, 0)
    gw_waitbar<W_P3>();
    GW_MFMA(0, 1)
  }

  {  // peeled last tile
    const int buf = (nkt - 1) & 1;
    GW_READ_A(buf, 0)
    GW_READ_B(buf, 0)
    gw_waitbar<P_P0>();
    GW_MFMA(0, 0)
    GW_READ_A(buf, 1)
    gw_waitbar<P_P1>();
    GW_MFMA(1, 0)
    GW_READ_B(buf, 1)
    GW_BAR()
    GW_MFMA(1, 1)
    GW_READ_A(buf, 0)
    GW_MFMA(0, 1)
  }

#undef GW_READ_A
#undef GW_READ_B
#undef GW_MFMA
#undef GW_BAR

  const int col0 = tile_n * BN + wc * 64 + fr;
  const long row_base = tile_m * BM + wr * WM + (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 2 * MF; ++m) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long row = row_base + m * 16 + rg;
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        float v = acc[m][n][rg];
        short* dp2 = &D[row * N + col0 + n * 16];
        if constexpr (ACC) v += float(*reinterpret_cast<c10::BFloat16*>(dp2));
        *dp2 = (short)c10::BFloat16(v).x;
      }
    }
  }
  if (SP != nullptr) {
    __syncthreads();
    stats_epilogue<MF, WAVES_M, WAVES_N>(SP, lds, acc, lane, wave, wr,
                                         wc, tile_n, N,
                                         (unsigned)blockIdx.x);
  } else if (BNA.partials != nullptr) {
    __syncthreads();
    bnbwd_epilogue<MF, WAVES_M, WAVES_N>(BNA, lds, acc, lane, wave, wr,
                                         wc, tile_m * 256L, tile_n, N,
                                         (unsigned)blockIdx.x);
  }
}

}  // namespace

torch::Tensor gemm_btw_bnbwd_bf16(torch::Tensor a, torch::Tensor b,
                                  torch::Tensor bn_x,
                                  torch::Tensor bn_mean,
                                  torch::Tensor bn_rstd,
                                  torch::Tensor bn_gamma,
                                  torch::Tensor bn_beta,
                                  torch::Tensor bn_partials);

torch::Tensor gemm_btw_bf16(torch::Tensor a, torch::Tensor b,
                            torch::Tensor stats) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16
              && b.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1));
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(M % 256 == 0 && N % 64 == 0 && K % 64 == 0,
              "gemm_btw_bf16 requires M%256==0, N%64==0, K%64==0");
  auto d = torch::empty({M, N}, a.options());
  auto stream = c10::hip::getCurrentHIPStream();
  float* sp = nullptr;
  if (stats.defined() && stats.numel() > 0) {
    TORCH_CHECK(stats.scalar_type() == at::kFloat && stats.is_cuda()
                && stats.numel() == 16 * 2 * (long)N
                && stats.is_contiguous(),
                "stats must be zeroed float32 [16, 2*N]");
    sp = stats.data_ptr<float>();
  }
  const short* ap = (const short*)a.data_ptr();
  const short* bp = (const short*)b.data_ptr();
  short* dp = (short*)d.data_ptr();
  BnBwdArgs bna = {nullptr, nullptr, nullptr, nullptr, nullptr,
                   nullptr, nullptr};
  if (N % 128 == 0) {
    const long grid = ((long)M / 256) * (N / 128);
    hipLaunchKernelGGL((gemm_btw_kernel<64, 4, 2>), dim3((unsigned)grid),
                       dim3(GW_THREADS), 0, stream.stream(), ap, bp, dp,
                       sp, bna, M, N, K);
  } else {
    const long grid = ((long)M / 256) * (N / 64);
    hipLaunchKernelGGL((gemm_btw_kernel<32, 8, 1>), dim3((unsigned)grid),
                       dim3(GW_THREADS), 0, stream.stream(), ap, bp, dp,
                       sp, bna, M, N, K);
  }
  return d;
}

torch::Tensor gemm_btw_bnbwd_bf16(torch::Tensor a, torch::Tensor b,
                                  torch::Tensor bn_x,
                                  torch::Tensor bn_mean,
                                  torch::Tensor bn_rstd,
                                  torch::Tensor bn_gamma,
                                  torch::Tensor bn_beta,
                                  torch::Tensor bn_partials) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16
              && b.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1));
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(M % 256 == 0 && N % 64 == 0 && K % 64 == 0);
  TORCH_CHECK(bn_partials.scalar_type() == at::kFloat
              && bn_partials.numel() == 16 * 2 * (long)N
              && bn_partials.is_contiguous());
  TORCH_CHECK(bn_x.scalar_type() == at::kBFloat16
              && bn_x.numel() == (long)M * N);
  TORCH_CHECK(bn_mean.numel() == N && bn_rstd.numel() == N
              && bn_gamma.numel() == N && bn_beta.numel() == N);
  auto d = torch::empty({M, N}, a.options());
  auto stream = c10::hip::getCurrentHIPStream();
  const short* ap = (const short*)a.data_ptr();
  const short* bp = (const short*)b.data_ptr();
  short* dp = (short*)d.data_ptr();
  BnBwdArgs bna = {(const short*)bn_x.data_ptr(),
                   bn_mean.data_ptr<float>(), bn_rstd.data_ptr<float>(),
                   (const short*)bn_gamma.data_ptr(),
                   (const short*)bn_beta.data_ptr(),
                   bn_partials.data_ptr<float>(), (const short*)nullptr};
  bna.d = dp;
  if (N % 128 == 0) {
    const long grid = ((long)M / 256) * (N / 128);
    hipLaunchKernelGGL((gemm_btw_kernel<64, 4, 2>), dim3((unsigned)grid),
                       dim3(GW_THREADS), 0, stream.stream(), ap, bp, dp,
                       nullptr, bna, M, N, K);
  } else {
    const long grid = ((long)M / 256) * (N / 64);
    hipLaunchKernelGGL((gemm_btw_kernel<32, 8, 1>), dim3((unsigned)grid),
                       dim3(GW_THREADS), 0, stream.stream(), ap, bp, dp,
                       nullptr, bna, M, N, K);
  }
  return d;
}

torch::Tensor gemm_btw_acc_bf16(torch::Tensor a, torch::Tensor b,
                                torch::Tensor out) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16
              && b.scalar_type() == at::kBFloat16
              && out.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1));
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous()
              && out.is_contiguous());
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(out.size(0) == M && out.size(1) == N);
  TORCH_CHECK(M % 256 == 0 && N % 64 == 0 && K % 64 == 0);
  auto stream = c10::hip::getCurrentHIPStream();
  const short* ap = (const short*)a.data_ptr();
  const short* bp = (const short*)b.data_ptr();
  short* dp = (short*)out.data_ptr();
  if (N % 128 == 0) {
    const long grid = ((long)M / 256) * (N / 128);
    hipLaunchKernelGGL((gemm_btw_kernel<64, 4, 2, true>),
                       dim3((unsigned)grid), dim3(GW_THREADS), 0,
                       stream.stream(), ap, bp, dp, nullptr,
                       BnBwdArgs{nullptr, nullptr, nullptr, nullptr,
                                 nullptr, nullptr, nullptr},
                       M, N, K);
  } else {
    const long grid = ((long)M / 256) * (N / 64);
    hipLaunchKernelGGL((gemm_btw_kernel<32, 8, 1, true>),
                       dim3((unsigned)grid), dim3(GW_THREADS), 0,
                       stream.stream(), ap, bp, dp, nullptr,
                       BnBwdArgs{nullptr, nullptr, nullptr, nullptr,
                                 nullptr, nullptr, nullptr},
                       M, N, K);
  }
  return out;
}
