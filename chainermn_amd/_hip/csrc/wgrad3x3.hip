// 3x3/s1/p1 convolution WEIGHT gradient as implicit GEMM (gfx950),
// channels_last bf16, fp32 accumulation.
//
//   gw[co, ky, kx, ci] = sum_m gy[m, co] * x[m + shift(ky,kx), ci]
//   (m = (n, oh, ow) over N*H*W; out-of-image taps contribute 0)
//
// A GEMM whose REDUCTION dimension is m: both operands are m-major in
// memory ([m][channel]) while the MFMA wants K(m)-contiguous per-lane
// fragments.  The transpose happens during staging: each thread loads
// two 16-byte channel rows (m and m+1), interleaves them into eight
// 32-bit [m|m+1] pairs in registers, and writes a pair-major LDS image
// [pair_row][64 b32 cells] with a row-XOR cell swizzle (bank(b32) is
// cell-only — row*64 = 0 mod 32 — so same-cell/different-row writes
// would otherwise fully serialize).  A fragment's 8 m-values of one
// channel are then four ds_read_b32 (conflict-free: 16 distinct cells
// per group, same row).  The xi (im2col) operand masks out-of-image
// taps to zero at load (positions advance per tile with one magic
// division by W — no runtime division in the loop).
//
// Grid: (Co/64) x (9*Ci/64) x SPLIT blocks; each block accumulates
// D[64co x 64k] over its m-slice (waves split m within a tile, acc in
// registers across all tiles) and atomicAdds fp32 into the pre-zeroed
// gw buffer; the Python wrapper casts to bf16 [Co, Ci, 3, 3] CL.
//
// Replaces MIOpen's igemm_wrw kernels (CDNA2-era 32x32x8 MFMA) on
// dispatched shapes; per-shape autotune + fallback in ops.conv3x3_wgrad.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <algorithm>

#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define WG_THREADS 512
#define WG_MB 256  // m per tile (128 pair rows)

__device__ __forceinline__ unsigned mdiv(unsigned n, unsigned mul,
                                         int shift) {
  // q = floor(n / d) for n < 2^31 given (mul, shift) from wg_magic()
  return (unsigned)(((unsigned long long)n * mul) >> (32 + shift));
}

__device__ __forceinline__ int wg_swz(int pair_row, int cell) {
  return cell ^ ((pair_row & 15) << 1);
}

__global__ __launch_bounds__(WG_THREADS, 1) void wgrad3x3_kernel(
    const short* __restrict__ X,   // [N, H, W, Ci]
    const short* __restrict__ GY,  // [N, H, W, Co]
    float* __restrict__ GW,        // [Co, 9*Ci] fp32, pre-zeroed
    long M, int H, int Wd, int Ci, int Co,
    unsigned mg_hw, int sh_hw,  // magic divisor for H*W
    unsigned mg_w, int sh_w,    // magic divisor for W
    long m_per_split) {
  // one 128-pair x 64-cell b32 image per operand = 32 KiB each
  __shared__ __attribute__((aligned(16))) unsigned lds[2 * 128 * 64];
  unsigned* img_gy = lds;
  unsigned* img_xi = lds + 128 * 64;

  const int nk = 9 * Ci / 64;
  const int nco = Co / 64;
  const int bid = blockIdx.x;
  const int co_t = bid % nco;
  const int k_t = (bid / nco) % nk;
  const int split = bid / (nco * nk);

  const int tap = (k_t * 64) / Ci;  // Ci % 64 == 0: tile inside one tap
  const int ky = tap / 3, kx = tap % 3;
  const int ci0 = k_t * 64 - tap * Ci;

  const long m_begin = (long)split * m_per_split;
  const long m_end = min(m_begin + m_per_split, M);

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int HWl = H * Wd;

  // staging: 1024 pair-units per tile (128 pair rows x 8 ch octets);
  // thread owns units t and t+512: p = u>>3, g = u&7
  int u_n[2], u_pos[2];  // image index / position-in-image of row m0
  const int adv_n = (int)((unsigned)WG_MB / (unsigned)HWl);
  const int adv_r = (int)((unsigned)WG_MB % (unsigned)HWl);
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int u = i * 512 + t;
    const unsigned mu = (unsigned)(m_begin + 2 * (long)(u >> 3));
    const unsigned n = mdiv(mu, mg_hw, sh_hw);
    u_n[i] = (int)n;
    u_pos[i] = (int)(mu - n * (unsigned)HWl);
  }

  f32x4 acc[4][4];  // [co frag][k frag]
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 4; ++b) acc[a][b] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int fr = lane & 15;
  const int kq = lane >> 4;

  const long ntiles = (m_end - m_begin + WG_MB - 1) / WG_MB;

  // ---- T14 async-stage split: issue tile t+1's GLOBAL loads before
  // compute(t) so HBM latency hides under the MFMA cluster; the
  // register->LDS write pass lands after the barrier ----
  uint4 ga[2], gb[2], xa[2], xb[2];  // in-flight rows per unit

  auto issue_loads = [&](long mtile) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int u = i * 512 + t;
      const int p = u >> 3;
      const int g = u & 7;
      const long gm0 = mtile + 2 * (long)p;
      const bool va = gm0 < m_end;
      const bool vb = gm0 + 1 < m_end;
      ga[i] = (uint4){0, 0, 0, 0};
      gb[i] = (uint4){0, 0, 0, 0};
      xa[i] = (uint4){0, 0, 0, 0};
      xb[i] = (uint4){0, 0, 0, 0};
      if (va)
        ga[i] = *reinterpret_cast<const uint4*>(
            GY + gm0 * Co + co_t * 64 + g * 8);
      if (vb)
        gb[i] = *reinterpret_cast<const uint4*>(
            GY + (gm0 + 1) * Co + co_t * 64 + g * 8);
      const int pos_a = u_pos[i];
      const int n_a = u_n[i];
      int pos_b = pos_a + 1, n_b = n_a;
      if (pos_b >= HWl) { pos_b = 0; n_b += 1; }
      const int oh_a = (int)mdiv((unsigned)pos_a, mg_w, sh_w);
      const int ow_a = pos_a - oh_a * Wd;
      const int oh_b = (int)mdiv((unsigned)pos_b, mg_w, sh_w);
      const int ow_b = pos_b - oh_b * Wd;
      const int ih_a = oh_a + ky - 1, iw_a = ow_a + kx - 1;
      const int ih_b = oh_b + ky - 1, iw_b = ow_b + kx - 1;
      if (va && (unsigned)ih_a < (unsigned)H
          && (unsigned)iw_a < (unsigned)Wd)
        xa[i] = *reinterpret_cast<const uint4*>(
            X + (((long)n_a * H + ih_a) * Wd + iw_a) * Ci + ci0 + g * 8);
      if (vb && (unsigned)ih_b < (unsigned)H
          && (unsigned)iw_b < (unsigned)Wd)
        xb[i] = *reinterpret_cast<const uint4*>(
            X + (((long)n_b * H + ih_b) * Wd + iw_b) * Ci + ci0 + g * 8);
      // advance position state to the NEXT tile
      u_pos[i] += adv_r;
      u_n[i] += adv_n;
      if (u_pos[i] >= HWl) { u_pos[i] -= HWl; u_n[i] += 1; }
    }
  };

  auto write_tiles = [&]() {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int u = i * 512 + t;
      const int p = u >> 3;
      const int g = u & 7;
      const unsigned* pa = reinterpret_cast<const unsigned*>(&ga[i]);
      const unsigned* pb = reinterpret_cast<const unsigned*>(&gb[i]);
      const unsigned* qa = reinterpret_cast<const unsigned*>(&xa[i]);
      const unsigned* qb = reinterpret_cast<const unsigned*>(&xb[i]);
      unsigned* gi = img_gy + p * 64;
      unsigned* xi = img_xi + p * 64;
#pragma unroll
      for (int d = 0; d < 4; ++d) {
        const int c0 = g * 8 + d * 2;
        gi[wg_swz(p, c0)] = (pa[d] & 0xFFFFu) | (pb[d] << 16);
        gi[wg_swz(p, c0 + 1)] = (pa[d] >> 16) | (pb[d] & 0xFFFF0000u);
        xi[wg_swz(p, c0)] = (qa[d] & 0xFFFFu) | (qb[d] << 16);
        xi[wg_swz(p, c0 + 1)] = (qa[d] >> 16) | (qb[d] & 0xFFFF0000u);
      }
    }
  };

  auto compute = [&]() {
    const int prow0 = wave * 16 + kq * 4;
    bf16x8 af[4], bfr[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      unsigned vg[4], vx[4];
#pragma unroll
      for (int pp = 0; pp < 4; ++pp) {
        const int r = prow0 + pp;
        vg[pp] = img_gy[r * 64 + wg_swz(r, f * 16 + fr)];
        vx[pp] = img_xi[r * 64 + wg_swz(r, f * 16 + fr)];
      }
      af[f] = *reinterpret_cast<bf16x8*>(vg);
      bfr[f] = *reinterpret_cast<bf16x8*>(vx);
    }
#pragma unroll
    for (int a = 0; a < 4; ++a)
#pragma unroll
      for (int b = 0; b < 4; ++b)
        acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[a], bfr[b], acc[a][b], 0, 0, 0);
  };

  if (ntiles > 0) {
    issue_loads(m_begin);
    write_tiles();
    __syncthreads();
  }
  for (long tt = 0; tt < ntiles; ++tt) {
    if (tt + 1 < ntiles)
      issue_loads(m_begin + (tt + 1) * WG_MB);  // in flight under MFMA
    compute();
    __syncthreads();  // everyone done reading the images
    if (tt + 1 < ntiles) {
      write_tiles();
      __syncthreads();  // images ready for the next compute
    }
  }

  // ---- epilogue: acc col = k (lane&15), row = co ((lane>>4)*4 + reg) ----
  const long kcol0 = (long)k_t * 64 + fr;
  const int corow0 = co_t * 64 + kq * 4;
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long co = corow0 + a * 16 + rg;
#pragma unroll
      for (int b = 0; b < 4; ++b)
        atomicAdd(&GW[co * (9 * (long)Ci) + kcol0 + b * 16],
                  acc[a][b][rg]);
    }
}

struct WgMagic {
  unsigned mul;
  int shift;
};

WgMagic wg_magic(unsigned d, unsigned long long n_max) {
  // (mul, shift) with floor(n/d) == (n*mul) >> (32+shift) for n <= n_max.
  // Round-up magic m = ceil(2^(32+s)/d) is exact for n < 2^(32+s)/e with
  // e = m*d - 2^(32+s); pick the largest s whose m fits 32 bits and
  // whose exactness bound covers n_max (verified in tests).
  if ((d & (d - 1)) == 0) {  // power of two (incl. 1): pure shift
    int k = 0;
    while ((1u << k) < d) ++k;
    return {1u << 31, k - 1};
  }
  for (int s = 30; s >= 0; --s) {
    const unsigned long long m = ((1ULL << (32 + s)) + d - 1) / d;
    if (m >= (1ULL << 32)) continue;
    const unsigned long long e = m * d - (1ULL << (32 + s));
    if (e == 0 || (1ULL << (32 + s)) / e > n_max)
      return {(unsigned)m, s};
  }
  TORCH_CHECK(false, "no exact magic divisor for d=", d);
}

}  // namespace

torch::Tensor conv3x3_wgrad_bf16(torch::Tensor x, torch::Tensor gy) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16
              && gy.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.dim() == 4
              && x.is_contiguous(at::MemoryFormat::ChannelsLast)
              && gy.is_contiguous(at::MemoryFormat::ChannelsLast));
  const long Nb = x.size(0);
  const int Ci = (int)x.size(1), H = (int)x.size(2), Wd = (int)x.size(3);
  const int Co = (int)gy.size(1);
  TORCH_CHECK(gy.size(0) == Nb && gy.size(2) == H && gy.size(3) == Wd);
  TORCH_CHECK(Ci % 64 == 0 && Co % 64 == 0,
              "requires Ci%64==0 and Co%64==0");
  const long M = Nb * (long)H * Wd;
  TORCH_CHECK(M < (1L << 31), "N*H*W must fit in 32 bits");
  TORCH_CHECK(H * Wd >= 2, "degenerate spatial");
  auto gw32 = torch::zeros({(long)Co, 9L * Ci},
                           x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  const int nco = Co / 64, nk = 9 * Ci / 64;
  // SPLIT: ~2048 blocks (2 resident per CU = 512 concurrent slots; 4x
  // oversubscription amortizes the ragged last dispatch wave), m-slices
  // in whole 256-m tiles
  int split = (int)std::min<long>(
      std::max<long>(1, 2048 / ((long)nco * nk)),
      std::max<long>(1, (M + WG_MB - 1) / WG_MB));
  long tiles = (M + WG_MB - 1) / WG_MB;
  long tiles_per_split = (tiles + split - 1) / split;
  long m_per_split = tiles_per_split * WG_MB;
  split = (int)((M + m_per_split - 1) / m_per_split);
  const WgMagic mhw = wg_magic((unsigned)(H * Wd), (unsigned)M);
  const WgMagic mw = wg_magic((unsigned)Wd, (unsigned)(H * Wd));
  const long grid = (long)nco * nk * split;
  hipLaunchKernelGGL(wgrad3x3_kernel, dim3((unsigned)grid),
                     dim3(WG_THREADS), 0, stream.stream(),
                     (const short*)x.data_ptr(),
                     (const short*)gy.data_ptr(), gw32.data_ptr<float>(),
                     M, H, Wd, Ci, Co, mhw.mul, mhw.shift, mw.mul,
                     mw.shift, m_per_split);
  // [Co, 9*Ci] -> [Co, 3, 3, Ci] -> channels_last [Co, Ci, 3, 3] bf16
  auto gw = gw32.view({(long)Co, 3, 3, (long)Ci})
                .permute({0, 3, 1, 2})
                .to(at::kBFloat16)
                .contiguous(at::MemoryFormat::ChannelsLast);
  return gw;
}
