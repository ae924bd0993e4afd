// 256x256-tile MFMA bf16 GEMM (gfx950): D[M,N] = A[M,K] @ B[N,K]^T.
//
// The deep-schedule successor of gemm.hip's 128x128 kernel (which sits at
// the 2-barrier structure's ~900 TF ceiling).  Geometry per
// /opt/skills/guides/cdna_hip_programming.md §5 "256² template":
//
//  * 256x256 block tile, BK=64; 8 waves (512 threads) as 2(M)x4(N), each
//    wave owning a 128x64 output = 8x4 fragments of
//    v_mfma_f32_16x16x32_bf16 (128 f32 accumulator registers per lane)
//  * 2 LDS buffers x (A 256x64 + B 256x64) bf16 = 128 KiB of the CU's
//    160 KiB; staging via __builtin_amdgcn_global_load_lds width 16
//    (4 glds per matrix per K-tile)
//  * LDS bank-conflict XOR swizzle on the per-lane *source* address +
//    the ds_read_b128 offsets (lane-linear glds destination)
//  * XCD-aware bijective blockIdx remap
//
// Requires M%256 == N%256 == 0, K%64 == 0 (the Python wrapper autotunes
// this vs the 128-tile kernel vs hipBLASLt per shape).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define G2_BM 256
#define G2_BN 256
#define G2_BK 64
#define G2_THREADS 512

__device__ __forceinline__ int swz2(int row, int within_byte) {
  return within_byte ^ ((row & 7) << 4);
}

__global__ __launch_bounds__(G2_THREADS, 1) void gemm_bt256_bf16_kernel(
    const short* __restrict__ A,  // [M, K] bf16 bits
    const short* __restrict__ B,  // [N, K] bf16 bits
    short* __restrict__ D,        // [M, N] bf16 bits
    int M, int N, int K) {
  // [buf][mat][row][128-byte swizzled row]
  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * G2_BM * G2_BK * 2];
  auto ldsA = [&](int buf) -> char* {
    return lds + buf * 2 * G2_BM * G2_BK * 2;
  };
  auto ldsB = [&](int buf) -> char* {
    return lds + buf * 2 * G2_BM * G2_BK * 2 + G2_BM * G2_BK * 2;
  };

  const int nwg = gridDim.x;
  const int orig = blockIdx.x;
  const int q = nwg / 8, r = nwg % 8, xcd = orig % 8;
  const int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                   + orig / 8;
  const int ntile_n = N / G2_BN;
  const int tile_m = wgid / ntile_n;
  const int tile_n = wgid % ntile_n;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 2;   // wave row (0..1): 128 output rows each
  const int wc = wave & 3;    // wave col (0..3): 64 output cols each

  // ---- staging: 4 glds per wave per matrix per K-tile ----
  auto stage = [&](int buf, int kt) {
    const long k0 = (long)kt * G2_BK;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int p = (i * 8 + wave) * 1024 + lane * 16;
      const int row = p >> 7;
      const int within = p & 127;
      const short* srcA = A + ((long)(tile_m * G2_BM + row)) * K + k0
                          + (swz2(row, within) >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)
              (ldsA(buf) + (i * 8 + wave) * 1024),
          16, 0, 0);
      const short* srcB = B + ((long)(tile_n * G2_BN + row)) * K + k0
                          + (swz2(row, within) >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcB,
          (__attribute__((address_space(3))) unsigned int*)
              (ldsB(buf) + (i * 8 + wave) * 1024),
          16, 0, 0);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int m = 0; m < 8; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int nkt = K / G2_BK;
  stage(0, 0);
  __syncthreads();  // drains the in-flight LDS-DMA (vmcnt 0) + barrier

  const int fr = lane & 15;  // fragment row within 16
  const int kq = lane >> 4;  // k-quarter: 8 bf16 each

  for (int kt = 0; kt < nkt; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < nkt) stage(buf ^ 1, kt + 1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // two K=32 steps per K-tile
      const int kbyte = ks * 64 + kq * 16;
      bf16x8 b_frag[4];
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int row = wc * 64 + n * 16 + fr;
        b_frag[n] = *(const bf16x8*)(ldsB(buf) + row * 128
                                     + swz2(row, kbyte));
      }
#pragma unroll
      for (int m = 0; m < 8; ++m) {
        const int row = wr * 128 + m * 16 + fr;
        const bf16x8 a_frag = *(const bf16x8*)(ldsA(buf) + row * 128
                                               + swz2(row, kbyte));
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag[n], acc[m][n], 0, 0, 0);
      }
    }
    __syncthreads();  // next-tile DMA drained (vmcnt 0) + wave join
  }

  // ---- epilogue: C/D map for 16x16x32: col = lane&15,
  // row = (lane>>4)*4 + reg ----
  const int col0 = tile_n * G2_BN + wc * 64 + fr;
  const int row_base = tile_m * G2_BM + wr * 128 + (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 8; ++m) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long row = row_base + m * 16 + rg;
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const float v = acc[m][n][rg];
        D[row * N + col0 + n * 16] = (short)c10::BFloat16(v).x;
      }
    }
  }
}

// ---- 8-phase schedule (T3+T4+T5 of the guide) ---------------------------
//
// Each K-tile is split into 4 phases; per phase one output QUADRANT of the
// wave's 128x64 block (qa = which 64-row half, qb = which 32-col half) is
// computed over the full K=64, while ONE 16 KiB half-tile for K-tile t+1
// is staged (2 glds).  Phases walk quadrants in gray order
// (0,0)->(1,0)->(1,1)->(0,1) and stage halves in order A0,B0,A1,B1, so a
// counted s_waitcnt vmcnt(4) at the end of phases 0, 1 and 3 (none at 2)
// guarantees every half is resident 1+ phase before its first read while
// ALWAYS leaving 2 half-tiles (4 glds) in flight across the barriers —
// the pipeline never drains in the main loop.  Raw s_barrier (inside one
// asm statement with the wait, so compiler-visible LDS reads cannot hoist
// past it) instead of __syncthreads() keeps hipcc from inserting the
// vmcnt(0) drain that caps the 2-barrier structure.
// SCHED bit0: issue the half-tile stage BEFORE the ds_reads of the phase
// (T3 recipe order).  SCHED bit1: drop the barrier AFTER each MFMA
// cluster (1 barrier per phase): consecutive phases of the two
// co-resident waves can then pair a load segment against the partner's
// MFMA segment instead of matrix||matrix.
// ACC: accumulate into D (D += A@B^T) instead of overwriting — used to
// fold the residual-junction gradient add into the dgrad epilogue.
template <int SCHED, bool ACC = false>
__global__ __launch_bounds__(G2_THREADS, 1) void gemm_bt256p_bf16_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ D, float* __restrict__ SP, BnBwdArgs BNA,
    int M, int N, int K) {
  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * G2_BM * G2_BK * 2];
  auto ldsA = [&](int buf) -> char* {
    return lds + buf * 2 * G2_BM * G2_BK * 2;
  };
  auto ldsB = [&](int buf) -> char* {
    return lds + buf * 2 * G2_BM * G2_BK * 2 + G2_BM * G2_BK * 2;
  };

  const int nwg = gridDim.x;
  const int orig = blockIdx.x;
  const int q = nwg / 8, r = nwg % 8, xcd = orig % 8;
  const int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                   + orig / 8;
  const int ntile_n = N / G2_BN;
  const int tile_m = wgid / ntile_n;
  const int tile_n = wgid % ntile_n;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 2;
  const int wc = wave & 3;
  const int fr = lane & 15;
  const int kq = lane >> 4;

  // stage one 16 KiB half-tile: which = 0..3 -> A-half0, B-half0,
  // A-half1, B-half1.  A-half h = tile rows {h*64..h*64+63} u
  // {128+h*64..128+h*64+63} (the h-th 64-row half of BOTH wave-row
  // blocks); B-half h = rows {64k + h*32 .. +31, k=0..3}.  Each chunk is
  // 8 contiguous LDS rows = 1 KiB, lane-linear for glds.
  auto stage_half = [&](int buf, int kt, int which) {
    const long k0 = (long)kt * G2_BK;
    const int mat = which & 1;
    const int h = which >> 1;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int c = i * 8 + wave;  // chunk 0..15
      const int row0 = (mat == 0)
          ? (c < 8 ? h * 64 + c * 8 : 128 + h * 64 + (c - 8) * 8)
          : 64 * (c >> 2) + h * 32 + (c & 3) * 8;
      const int p = lane * 16;
      const int row = row0 + (p >> 7);
      const int within = p & 127;
      const short* src = (mat == 0
          ? A + ((long)(tile_m * G2_BM + row)) * K
          : B + ((long)(tile_n * G2_BN + row)) * K)
          + k0 + (swz2(row, within) >> 1);
      char* dst = (mat == 0 ? ldsA(buf) : ldsB(buf)) + row0 * 128;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int m = 0; m < 8; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  bf16x8 af[2][4];  // current A quadrant frags [ks][mi]
  bf16x8 bfr[2][2];  // current B quadrant frags [ks][ni]

#define G2_READ_A(buf, qa)                                                   \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int mi = 0; mi < 4; ++mi) {                     \
    const int row = wr * 128 + (qa) * 64 + mi * 16 + fr;                     \
    af[ks][mi] = *(const bf16x8*)(ldsA(buf) + row * 128                      \
                                  + swz2(row, ks * 64 + kq * 16));           \
  }
#define G2_READ_B(buf, qb)                                                   \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int ni = 0; ni < 2; ++ni) {                     \
    const int row = wc * 64 + (qb) * 32 + ni * 16 + fr;                      \
    bfr[ks][ni] = *(const bf16x8*)(ldsB(buf) + row * 128                     \
                                   + swz2(row, ks * 64 + kq * 16));          \
  }
#define G2_MFMA(qa, qb)                                                      \
  __builtin_amdgcn_s_setprio(1);                                            \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                           \
      _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                       \
          _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                   \
    acc[(qa) * 4 + mi][(qb) * 2 + ni] =                                      \
        __builtin_amdgcn_mfma_f32_16x16x32_bf16(                             \
            af[ks][mi], bfr[ks][ni], acc[(qa) * 4 + mi][(qb) * 2 + ni],      \
            0, 0, 0);                                                        \
  __builtin_amdgcn_s_setprio(0);
#define G2_WAITBAR(n)                                                        \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_barrier" ::: "memory");
#define G2_BAR() asm volatile("s_barrier" ::: "memory");

#define G2_TRAILBAR()                                                        \
  if constexpr (!(SCHED & 2)) { G2_BAR() }

  const int nkt = K / G2_BK;
  // prologue: all 4 halves of tile 0; A0/B0 must land before phase 0
  stage_half(0, 0, 0);
  stage_half(0, 0, 1);
  stage_half(0, 0, 2);
  stage_half(0, 0, 3);
  G2_WAITBAR(4)

  for (int kt = 0; kt < nkt - 1; ++kt) {
    const int buf = kt & 1;
    const int nxt = buf ^ 1;
    // phase 0: quadrant (0,0); stage A-half0 of kt+1
    if constexpr (SCHED & 1) {
      stage_half(nxt, kt + 1, 0);
      G2_READ_A(buf, 0)
      G2_READ_B(buf, 0)
    } else {
      G2_READ_A(buf, 0)
      G2_READ_B(buf, 0)
      stage_half(nxt, kt + 1, 0);
    }
    G2_WAITBAR(4)  // guard A-half1(kt): leaves B1(kt), A0(kt+1) in flight
    G2_MFMA(0, 0)
    G2_TRAILBAR()
    // phase 1: quadrant (1,0) — B frags reused; stage B-half0 of kt+1
    if constexpr (SCHED & 1) {
      stage_half(nxt, kt + 1, 1);
      G2_READ_A(buf, 1)
    } else {
      G2_READ_A(buf, 1)
      stage_half(nxt, kt + 1, 1);
    }
    G2_WAITBAR(4)  // guard B-half1(kt): leaves A0/B0(kt+1) in flight
    G2_MFMA(1, 0)
    G2_TRAILBAR()
    // phase 2: quadrant (1,1) — A frags reused; stage A-half1 of kt+1
    if constexpr (SCHED & 1) {
      stage_half(nxt, kt + 1, 2);
      G2_READ_B(buf, 1)
    } else {
      G2_READ_B(buf, 1)
      stage_half(nxt, kt + 1, 2);
    }
    G2_BAR()  // A-half0(kt) still resident: no wait needed
    G2_MFMA(1, 1)
    G2_TRAILBAR()
    // phase 3: quadrant (0,1) — B frags reused; stage B-half1 of kt+1
    if constexpr (SCHED & 1) {
      stage_half(nxt, kt + 1, 3);
      G2_READ_A(buf, 0)
    } else {
      G2_READ_A(buf, 0)
      stage_half(nxt, kt + 1, 3);
    }
    G2_WAITBAR(4)  // guard A0/B0(kt+1): leaves A1/B1(kt+1) in flight
    G2_MFMA(0, 1)
    G2_TRAILBAR()
  }

  {  // peeled last tile: nothing to stage; drain 4 -> 2 -> 0
    const int buf = (nkt - 1) & 1;
    G2_READ_A(buf, 0)
    G2_READ_B(buf, 0)
    G2_WAITBAR(2)  // guard A-half1
    G2_MFMA(0, 0)
    G2_TRAILBAR()
    G2_READ_A(buf, 1)
    G2_WAITBAR(0)  // guard B-half1
    G2_MFMA(1, 0)
    G2_TRAILBAR()
    G2_READ_B(buf, 1)
    G2_BAR()
    G2_MFMA(1, 1)
    G2_TRAILBAR()
    G2_READ_A(buf, 0)
    G2_MFMA(0, 1)
  }
#undef G2_TRAILBAR

#undef G2_READ_A
#undef G2_READ_B
#undef G2_MFMA
#undef G2_WAITBAR
#undef G2_BAR

  const int col0 = tile_n * G2_BN + wc * 64 + fr;
  const int row_base = tile_m * G2_BM + wr * 128 + (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 8; ++m) {
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
    stats_epilogue<4, 2, 4>(SP, lds, acc, lane, wave, wr, wc, tile_n, N,
                            (unsigned)blockIdx.x);
  } else if (BNA.partials != nullptr) {
    __syncthreads();
    bnbwd_epilogue<4, 2, 4>(BNA, lds, acc, lane, wave, wr, wc,
                            (long)tile_m * 256, tile_n, N,
                            (unsigned)blockIdx.x);
  }
}

}  // namespace

static torch::Tensor launch_gemm256(torch::Tensor a, torch::Tensor b,
                                    int phased_sched,
                                    torch::Tensor stats = {},
                                    BnBwdArgs bna = {nullptr, nullptr,
                                                     nullptr, nullptr,
                                                     nullptr, nullptr,
                                                     nullptr}) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16
              && b.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1));
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(M % G2_BM == 0 && N % G2_BN == 0 && K % G2_BK == 0,
              "gemm_bt256_bf16 requires M%256==N%256==0, K%64==0");
  auto d = torch::empty({M, N}, a.options());
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = (M / G2_BM) * (N / G2_BN);
  const short* ap = (const short*)a.data_ptr();
  const short* bp = (const short*)b.data_ptr();
  short* dp = (short*)d.data_ptr();
  bna.d = dp;
  float* sp = nullptr;
  if (stats.defined() && stats.numel() > 0) {
    TORCH_CHECK(phased_sched > 0, "stats need the phased kernel");
    TORCH_CHECK(stats.scalar_type() == at::kFloat && stats.is_cuda()
                && stats.numel() == 16 * 2 * (long)N
                && stats.is_contiguous(),
                "stats must be zeroed float32 [16, 2*N]");
    sp = stats.data_ptr<float>();
  }
  switch (phased_sched) {
    case 0:
      hipLaunchKernelGGL(gemm_bt256_bf16_kernel, dim3(grid),
                         dim3(G2_THREADS), 0, stream.stream(), ap, bp, dp,
                         M, N, K);
      break;
    case 1:
      hipLaunchKernelGGL(gemm_bt256p_bf16_kernel<0>, dim3(grid),
                         dim3(G2_THREADS), 0, stream.stream(), ap, bp, dp,
                         sp, bna, M, N, K);
      break;
    case 2:
      hipLaunchKernelGGL(gemm_bt256p_bf16_kernel<1>, dim3(grid),
                         dim3(G2_THREADS), 0, stream.stream(), ap, bp, dp,
                         sp, bna, M, N, K);
      break;
    case 3:
      hipLaunchKernelGGL(gemm_bt256p_bf16_kernel<2>, dim3(grid),
                         dim3(G2_THREADS), 0, stream.stream(), ap, bp, dp,
                         sp, bna, M, N, K);
      break;
    default:
      hipLaunchKernelGGL(gemm_bt256p_bf16_kernel<3>, dim3(grid),
                         dim3(G2_THREADS), 0, stream.stream(), ap, bp, dp,
                         sp, bna, M, N, K);
  }
  return d;
}

torch::Tensor gemm_bt256_bf16(torch::Tensor a, torch::Tensor b) {
  return launch_gemm256(a, b, 0);
}

torch::Tensor gemm_bt256p_bf16(torch::Tensor a, torch::Tensor b,
                               long sched) {
  return launch_gemm256(a, b, 1 + (int)sched);
}

torch::Tensor gemm_bt256p_stats_bf16(torch::Tensor a, torch::Tensor b,
                                     torch::Tensor stats, long sched) {
  return launch_gemm256(a, b, 1 + (int)sched, stats);
}

// out += a @ b^T in one pass (the residual-junction fold).  out must be
// a contiguous bf16 [M, N] view of pre-existing gradient memory.
torch::Tensor gemm_bt256p_acc_bf16(torch::Tensor a, torch::Tensor b,
                                   torch::Tensor out) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16
              && b.scalar_type() == at::kBFloat16
              && out.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1));
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous()
              && out.is_contiguous());
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(out.size(0) == M && out.size(1) == N);
  TORCH_CHECK(M % G2_BM == 0 && N % G2_BN == 0 && K % G2_BK == 0);
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = (M / G2_BM) * (N / G2_BN);
  BnBwdArgs bna0 = {nullptr, nullptr, nullptr, nullptr, nullptr,
                    nullptr, nullptr};
  hipLaunchKernelGGL((gemm_bt256p_bf16_kernel<3, true>), dim3(grid),
                     dim3(G2_THREADS), 0, stream.stream(),
                     (const short*)a.data_ptr(), (const short*)b.data_ptr(),
                     (short*)out.data_ptr(), nullptr, bna0, M, N, K);
  return out;
}

// D = A @ B^T with the BN-backward reduce of the produced gy fused into
// the epilogue (see conv3x3.hip's bnbwd entry for semantics).
torch::Tensor gemm_bt256p_bnbwd_bf16(
    torch::Tensor a, torch::Tensor b, torch::Tensor bn_x,
    torch::Tensor bn_mean, torch::Tensor bn_rstd, torch::Tensor bn_gamma,
    torch::Tensor bn_beta, torch::Tensor bn_partials) {
  const long N = b.size(0);
  TORCH_CHECK(bn_partials.scalar_type() == at::kFloat
              && bn_partials.numel() == 16 * 2 * N
              && bn_partials.is_contiguous());
  TORCH_CHECK(bn_x.scalar_type() == at::kBFloat16
              && bn_x.numel() == a.size(0) * N);
  TORCH_CHECK(bn_mean.numel() == N && bn_rstd.numel() == N
              && bn_gamma.numel() == N && bn_beta.numel() == N);
  BnBwdArgs bna = {(const short*)bn_x.data_ptr(),
                   bn_mean.data_ptr<float>(), bn_rstd.data_ptr<float>(),
                   (const short*)bn_gamma.data_ptr(),
                   (const short*)bn_beta.data_ptr(),
                   bn_partials.data_ptr<float>(), nullptr};
  return launch_gemm256(a, b, 1 + 3, {}, bna);
}
