// 3x3 / stride-1 / pad-1 convolution as implicit GEMM on gfx950 MFMA,
// channels_last — the round-2 production kernel (fwd + dgrad; wgrad
// stays on MIOpen's split-K wrw, see links/fused_conv.py).
//
//   y[m, co] = sum_{ky,kx,ci} x[n, oh+ky-1, ow+kx-1, ci] * w[co,ky,kx,ci]
//   M = N*H*W (m -> n,oh,ow),  K = 9*Cin (k -> (ky,kx), ci)
//
// Structure = gemm256.hip's winning phased schedule (256-row M tile,
// 8 waves, double-buffered LDS via global_load_lds, counted vmcnt at
// phase ends, one raw barrier per phase, s_setprio MFMA clusters),
// specialized per Cout:
//
//   Cout >= 256: BN=256, waves 2(M)x4(N), per-wave 128x64  (16 MFMA/ph)
//   Cout == 128: BN=128, waves 4x2,       per-wave  64x64  ( 8 MFMA/ph)
//   Cout ==  64: BN= 64, waves 8x1,       per-wave  32x64; runs the
//                2-phase loop (8 MFMA + 1 barrier per phase) — the
//                4-phase schedule left this geometry wait-bound
//
// The A operand (im2col view of x) is staged by global_load_lds with the
// PER-LANE SOURCE address doing the im2col: each lane's (n,oh,ow) base
// address and bounds are precomputed once per kernel (the output
// position set of a chunk never changes across K-tiles; only the tap
// offset does), and out-of-bounds taps redirect the lane's source to a
// zero-filled page — no branches, no register staging, and the XOR bank
// swizzle rides on the source address as in gemm256.hip.  Cin % 64 == 0
// keeps every 64-wide K-tile inside one (ky,kx) tap; tap/channel
// counters advance incrementally (no divisions in the loop).
//
// dgrad reuses this kernel: gx = conv3x3s1p1(gy, W^T with taps flipped)
// — the Python side pre-transposes w once per backward (see
// links/fused_conv.py), so both directions run the same MFMA path.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define C3_THREADS 512

__device__ __forceinline__ int c3_swz(int row, int within_byte) {
  return within_byte ^ ((row & 7) << 4);
}

template <int N>
__device__ __forceinline__ void c3_waitbar() {
  // N = outstanding glds ALLOWED across this barrier; N < 0: no wait.
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

// WM = per-wave output rows; WAVES_M x WAVES_N = 8 waves; BN = 64*WAVES_N.
template <int WM, int WAVES_M, int WAVES_N>
__global__ __launch_bounds__(C3_THREADS, 1) void conv3x3_igemm_kernel(
    const short* __restrict__ X,   // [N, H, W, Cin] bf16 bits
    const short* __restrict__ Wt,  // [Cout, 3, 3, Cin]
    const short* __restrict__ ZP,  // 128-byte zero page
    short* __restrict__ Y,         // [N, H, W, Cout]
    float* __restrict__ SP,        // fwd BN stats partials or nullptr
    BnBwdArgs BNA,                 // fused BN-bwd reduce (dgrad) or nulls
    int HW, int H, int Wd, int Cin, int Cout) {
  constexpr int BM = 256;
  constexpr int BN = 64 * WAVES_N;
  constexpr int MF = WM / 32;       // m-frags per quadrant
  // stage sizes in glds instructions per thread
  constexpr int SA = 2;                          // one A half = 16 KiB
  constexpr int SB = (WAVES_N >= 2) ? WAVES_N / 2 : 1;  // B half (or both)
  constexpr bool B_BOTH = (WAVES_N == 1);  // stage both B halves at p1
  // steady-state vmcnt allowances (see scheduling note in gemm256.hip;
  // derived from the per-config stage sizes):
  constexpr int W_P0 = B_BOTH ? SA : (SB + SA);      // guard A1(t)
  constexpr int W_P1 = B_BOTH ? -1 : (SA + SB);      // guard B1(t)
  constexpr int W_P3 = B_BOTH ? SA : (SA + SB);      // guard A0/B0(t+1)
  // peeled tile allowances
  constexpr int P_P0 = B_BOTH ? 0 : SB;              // guard A1(last)
  constexpr int P_P1 = B_BOTH ? -1 : 0;              // guard B1(last)

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
  const int ntile_n = Cout / BN;
  const long tile_m = wgid / ntile_n;
  const int tile_n = wgid % ntile_n;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave / WAVES_N;
  const int wc = wave % WAVES_N;
  const int fr = lane & 15;
  const int kq = lane >> 4;

  // ---- per-thread A (im2col) stage descriptors: for (h, i) the thread
  // loads 16 B of chunk c = i*8+wave of A-half h; the output position m
  // for that 16 B is fixed for the whole kernel ----
  long a_base[2][2];   // X + m*Cin + swizzled element offset
  int a_oh[2][2], a_ow[2][2];
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int c = i * 8 + wave;
      constexpr int CPW = WM / 16;  // chunks per wave-row block per half
      const int row0 = (c / CPW) * WM + h * (WM / 2) + (c % CPW) * 8;
      const int p = lane * 16;
      const int row = row0 + (p >> 7);
      const int within = p & 127;
      const long m = tile_m * BM + row;
      const unsigned mu = (unsigned)m;
      const unsigned n = mu / (unsigned)HW;
      const unsigned rem = mu - n * (unsigned)HW;
      const int oh = (int)(rem / (unsigned)Wd);
      const int ow = (int)(rem - (unsigned)oh * (unsigned)Wd);
      a_oh[h][i] = oh;
      a_ow[h][i] = ow;
      a_base[h][i] = m * Cin + (c3_swz(row, within) >> 1);
    }

  // stage A-half h of K-tile at (tap ky,kx, channel block ci0)
  auto stage_a = [&](int buf, int h, int ky, int kx, int ci0) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int c = i * 8 + wave;
      constexpr int CPW = WM / 16;
      const int row0 = (c / CPW) * WM + h * (WM / 2) + (c % CPW) * 8;
      const int ih = a_oh[h][i] + ky - 1;
      const int iw = a_ow[h][i] + kx - 1;
      const bool ok = (unsigned)ih < (unsigned)H
                      && (unsigned)iw < (unsigned)Wd;
      const long off = a_base[h][i]
                       + (long)((ky - 1) * Wd + (kx - 1)) * Cin + ci0;
      const short* src = ok ? X + off : ZP;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              (ldsA(buf) + row0 * 128),
          16, 0, 0);
    }
  };

  // stage B (weights): half h (or both for WAVES_N==1); k0 = global k
  auto stage_b = [&](int buf, int h, int k0) {
    const int KW = 9 * Cin;
#pragma unroll
    for (int i = 0; i < SB; ++i) {
      int row0;
      if constexpr (B_BOTH) {
        row0 = wave * 8;
      } else {
        const int c = i * 8 + wave;  // SB*8 chunks over 8 waves
        row0 = (c / 4) * 64 + h * 32 + (c & 3) * 8;
      }
      const int p = lane * 16;
      const int row = row0 + (p >> 7);
      const int within = p & 127;
      const short* src = Wt + ((long)(tile_n * BN + row)) * KW + k0
                         + (c3_swz(row, within) >> 1);
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
  bf16x8 bfr1[2][2];  // second B half (B_BOTH 2-phase loop only)

#define C3_READ_A(buf, qa)                                                   \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int mi = 0; mi < MF; ++mi) {                    \
    const int row = wr * WM + (qa) * (WM / 2) + mi * 16 + fr;                \
    af[ks][mi] = *(const bf16x8*)(ldsA(buf) + row * 128                      \
                                  + c3_swz(row, ks * 64 + kq * 16));         \
  }
#define C3_READ_B(buf, qb)                                                   \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int ni = 0; ni < 2; ++ni) {                     \
    const int row = wc * 64 + (qb) * 32 + ni * 16 + fr;                      \
    bfr[ks][ni] = *(const bf16x8*)(ldsB(buf) + row * 128                     \
                                   + c3_swz(row, ks * 64 + kq * 16));        \
  }
#define C3_MFMA(qa, qb)                                                      \
  __builtin_amdgcn_s_setprio(1);                                            \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int mi = 0; mi < MF; ++mi)                      \
          _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                   \
    acc[(qa) * MF + mi][(qb) * 2 + ni] =                                     \
        __builtin_amdgcn_mfma_f32_16x16x32_bf16(                             \
            af[ks][mi], bfr[ks][ni], acc[(qa) * MF + mi][(qb) * 2 + ni],     \
            0, 0, 0);                                                        \
  __builtin_amdgcn_s_setprio(0);
#define C3_BAR() asm volatile("s_barrier" ::: "memory");

#define C3_READ_B1(buf)                                                     \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int ni = 0; ni < 2; ++ni) {                     \
    const int row = wc * 64 + 32 + ni * 16 + fr;                             \
    bfr1[ks][ni] = *(const bf16x8*)(ldsB(buf) + row * 128                    \
                                    + c3_swz(row, ks * 64 + kq * 16));       \
  }
#define C3_MFMA1(qa)                                                         \
  __builtin_amdgcn_s_setprio(1);                                            \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int mi = 0; mi < MF; ++mi)                      \
          _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                   \
    acc[(qa) * MF + mi][2 + ni] =                                            \
        __builtin_amdgcn_mfma_f32_16x16x32_bf16(                             \
            af[ks][mi], bfr1[ks][ni], acc[(qa) * MF + mi][2 + ni],           \
            0, 0, 0);                                                        \
  __builtin_amdgcn_s_setprio(0);

  const int nkt = 9 * (Cin / 64);
  // tap/channel counters for the tile being staged (one ahead); the
  // current tile's taps are never needed (its data is already in LDS)
  int sky = 0, skx = 0, sci0 = 0, sk0 = 0;
  auto advance = [&](int& ky_, int& kx_, int& ci0_) {
    ci0_ += 64;
    if (ci0_ == Cin) {
      ci0_ = 0;
      if (++kx_ == 3) { kx_ = 0; ++ky_; }
    }
  };

  if constexpr (B_BOTH) {
    // ---- 2-phase variant (Cout == 64): per phase 8 MFMAs instead of
    // 4, half the barriers — the 4-phase schedule left this geometry
    // wait-bound (8.2% MfmaUtil, profiles/conv3x3_pmc.txt).  Both B
    // halves stay live in registers; both waits are vmcnt(2).
    stage_a(0, 0, 0, 0, 0);
    stage_b(0, 0, 0);
    stage_a(0, 1, 0, 0, 0);
    c3_waitbar<2>();  // A0 + B01 landed; A1 in flight
    for (int kt = 0; kt < nkt - 1; ++kt) {
      const int buf = kt & 1;
      const int nxt = buf ^ 1;
      advance(sky, skx, sci0);
      sk0 += 64;
      // phase A: quadrants (0,0)+(0,1); stage A-half0(kt+1)
      stage_a(nxt, 0, sky, skx, sci0);
      C3_READ_A(buf, 0)
      C3_READ_B(buf, 0)
      C3_READ_B1(buf)
      c3_waitbar<2>();  // guard A1(kt); leaves A0(kt+1) in flight
      C3_MFMA(0, 0)
      C3_MFMA1(0)
      // phase B: quadrants (1,0)+(1,1); stage B then A-half1 of kt+1
      // (B BEFORE A1: the end-of-phase vmcnt(2) must leave only the A1
      // glds outstanding — B01 is consumed first next tile)
      stage_b(nxt, 0, sk0);
      stage_a(nxt, 1, sky, skx, sci0);
      C3_READ_A(buf, 1)
      c3_waitbar<2>();  // guard A0/B01(kt+1); leaves A1(kt+1) in flight
      C3_MFMA(1, 0)
      C3_MFMA1(1)
    }
    {  // peeled last tile
      const int buf = (nkt - 1) & 1;
      C3_READ_A(buf, 0)
      C3_READ_B(buf, 0)
      C3_READ_B1(buf)
      c3_waitbar<0>();  // guard A1(last)
      C3_MFMA(0, 0)
      C3_MFMA1(0)
      C3_READ_A(buf, 1)
      C3_BAR()
      C3_MFMA(1, 0)
      C3_MFMA1(1)
    }
  } else {
  // prologue: all halves of tile 0 in stage order (A0, B0, A1, B1)
    stage_a(0, 0, 0, 0, 0);
    stage_b(0, 0, 0);
    stage_a(0, 1, 0, 0, 0);
    if constexpr (!B_BOTH) stage_b(0, 1, 0);
    c3_waitbar<B_BOTH ? SA : SA + SB>();
  
    for (int kt = 0; kt < nkt - 1; ++kt) {
      const int buf = kt & 1;
      const int nxt = buf ^ 1;
      advance(sky, skx, sci0);
      sk0 += 64;
      // phase 0: quadrant (0,0); stage A-half0(kt+1)
      stage_a(nxt, 0, sky, skx, sci0);
      C3_READ_A(buf, 0)
      C3_READ_B(buf, 0)
      c3_waitbar<W_P0>();
      C3_MFMA(0, 0)
      // phase 1: quadrant (1,0); stage B-half0 (or both halves)
      stage_b(nxt, 0, sk0);
      C3_READ_A(buf, 1)
      c3_waitbar<W_P1>();
      C3_MFMA(1, 0)
      // phase 2: quadrant (1,1); stage A-half1
      stage_a(nxt, 1, sky, skx, sci0);
      C3_READ_B(buf, 1)
      C3_BAR()
      C3_MFMA(1, 1)
      // phase 3: quadrant (0,1); stage B-half1 (unless staged at p1)
      if constexpr (!B_BOTH) stage_b(nxt, 1, sk0);
      C3_READ_A(buf, 0)
      c3_waitbar<W_P3>();
      C3_MFMA(0, 1)
    }
  
    {  // peeled last tile
      const int buf = (nkt - 1) & 1;
      C3_READ_A(buf, 0)
      C3_READ_B(buf, 0)
      c3_waitbar<P_P0>();
      C3_MFMA(0, 0)
      C3_READ_A(buf, 1)
      c3_waitbar<P_P1>();
      C3_MFMA(1, 0)
      C3_READ_B(buf, 1)
      C3_BAR()
      C3_MFMA(1, 1)
      C3_READ_A(buf, 0)
      C3_MFMA(0, 1)
    }
  }

#undef C3_READ_B1
#undef C3_MFMA1
#undef C3_READ_A
#undef C3_READ_B
#undef C3_MFMA
#undef C3_BAR

  const int col0 = tile_n * BN + wc * 64 + fr;
  const long row_base = tile_m * BM + wr * WM + (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 2 * MF; ++m) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long row = row_base + m * 16 + rg;
#pragma unroll
      for (int n = 0; n < 4; ++n)
        Y[row * Cout + col0 + n * 16] =
            (short)c10::BFloat16(acc[m][n][rg]).x;
    }
  }
  if (SP != nullptr) {
    __syncthreads();  // all waves done with the main LDS buffers
    stats_epilogue<MF, WAVES_M, WAVES_N>(SP, lds, acc, lane, wave, wr,
                                         wc, tile_n, Cout,
                                         (unsigned)blockIdx.x);
  } else if (BNA.partials != nullptr) {
    __syncthreads();
    bnbwd_epilogue<MF, WAVES_M, WAVES_N>(BNA, lds, acc, lane, wave, wr,
                                         wc, tile_m * 256L, tile_n,
                                         Cout, (unsigned)blockIdx.x);
  }
}

torch::Tensor& zero_page(const torch::Tensor& like) {
  static torch::Tensor zp;
  if (!zp.defined() || zp.device() != like.device())
    zp = torch::zeros({64}, like.options().dtype(at::kBFloat16));
  return zp;
}

}  // namespace

torch::Tensor conv3x3s1_fwd_bnbwd_bf16(
    torch::Tensor x, torch::Tensor w, torch::Tensor stats,
    torch::Tensor bn_x, torch::Tensor bn_mean, torch::Tensor bn_rstd,
    torch::Tensor bn_gamma, torch::Tensor bn_beta,
    torch::Tensor bn_partials);

torch::Tensor conv3x3s1_fwd_bf16(torch::Tensor x, torch::Tensor w,
                                 torch::Tensor stats) {
  return conv3x3s1_fwd_bnbwd_bf16(x, w, stats, {}, {}, {}, {}, {}, {});
}

// dgrad form with the BN-backward reduce fused into the epilogue: the
// produced tensor is gy for the BN whose saved (x, mean, rstd, gamma,
// beta) are given; (sum g' | sum g'*xhat) partials land in bn_partials.
torch::Tensor conv3x3s1_fwd_bnbwd_bf16(
    torch::Tensor x, torch::Tensor w, torch::Tensor stats,
    torch::Tensor bn_x, torch::Tensor bn_mean, torch::Tensor bn_rstd,
    torch::Tensor bn_gamma, torch::Tensor bn_beta,
    torch::Tensor bn_partials) {
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
  TORCH_CHECK(Cin % 64 == 0 && Cout % 64 == 0,
              "requires Cin%64==0 and Cout%64==0");
  const long M = Nb * (long)H * Wd;
  TORCH_CHECK(M % 256 == 0, "requires N*H*W % 256 == 0");
  TORCH_CHECK(M < (1L << 31), "N*H*W must fit in 32 bits");
  auto y = torch::empty({Nb, Cout, H, Wd},
                        x.options().memory_format(
                            at::MemoryFormat::ChannelsLast));
  auto stream = c10::hip::getCurrentHIPStream();
  const short* xp = (const short*)x.data_ptr();
  const short* wp = (const short*)w.data_ptr();
  const short* zp = (const short*)zero_page(x).data_ptr();
  short* yp = (short*)y.data_ptr();
  float* sp = nullptr;
  if (stats.defined() && stats.numel() > 0) {
    TORCH_CHECK(stats.scalar_type() == at::kFloat && stats.is_cuda()
                && stats.numel() == 16 * 2 * (long)Cout
                && stats.is_contiguous(),
                "stats must be zeroed float32 [16, 2*Cout]");
    sp = stats.data_ptr<float>();
  }
  BnBwdArgs bna = {nullptr, nullptr, nullptr, nullptr, nullptr,
                   nullptr, nullptr};
  if (bn_partials.defined() && bn_partials.numel() > 0) {
    TORCH_CHECK(sp == nullptr, "stats and bn_partials are exclusive");
    TORCH_CHECK(bn_partials.scalar_type() == at::kFloat
                && bn_partials.numel() == 16 * 2 * (long)Cout
                && bn_partials.is_contiguous());
    TORCH_CHECK(bn_x.scalar_type() == at::kBFloat16
                && bn_x.is_contiguous(at::MemoryFormat::ChannelsLast)
                && bn_x.numel() == M * (long)Cout);
    TORCH_CHECK(bn_mean.numel() == Cout && bn_rstd.numel() == Cout
                && bn_gamma.numel() == Cout && bn_beta.numel() == Cout);
    bna = {(const short*)bn_x.data_ptr(), bn_mean.data_ptr<float>(),
           bn_rstd.data_ptr<float>(), (const short*)bn_gamma.data_ptr(),
           (const short*)bn_beta.data_ptr(),
           bn_partials.data_ptr<float>(), nullptr};
  }
  bna.d = yp;
  const int HW = H * Wd;
  if (Cout % 256 == 0) {
    const long grid = (M / 256) * (Cout / 256);
    hipLaunchKernelGGL((conv3x3_igemm_kernel<128, 2, 4>),
                       dim3((unsigned)grid), dim3(C3_THREADS), 0,
                       stream.stream(), xp, wp, zp, yp, sp, bna, HW, H,
                       Wd, Cin, Cout);
  } else if (Cout % 128 == 0) {
    const long grid = (M / 256) * (Cout / 128);
    hipLaunchKernelGGL((conv3x3_igemm_kernel<64, 4, 2>),
                       dim3((unsigned)grid), dim3(C3_THREADS), 0,
                       stream.stream(), xp, wp, zp, yp, sp, bna, HW, H,
                       Wd, Cin, Cout);
  } else {
    const long grid = (M / 256) * (Cout / 64);
    hipLaunchKernelGGL((conv3x3_igemm_kernel<32, 8, 1>),
                       dim3((unsigned)grid), dim3(C3_THREADS), 0,
                       stream.stream(), xp, wp, zp, yp, sp, bna, HW, H,
                       Wd, Cin, Cout);
  }
  return y;
}
