// Multi-node BatchNorm device kernels (gfx950).
//
// Rebuild of SURVEY.md §2.4 K6/K7: the reference computed local BN
// statistics with CuPy reductions and normalized with Chainer/cuDNN
// kernels.  Here:
//   bn_stats      : per-channel (Σx, Σx²) in ONE fused pass (the 2×C+1
//                   buffer that goes over RCCL), fp32 accumulation,
//                   wave-shuffle + LDS-tree partials, one atomicAdd per
//                   (block, channel)
//   bn_fwd        : y = (x-μ)·rstd·γ + β elementwise, reads the
//                   all-reduced stats
//   bn_bwd_reduce : per-channel (Σgy, Σgy·x̂) fused pass
//   bn_bwd_input  : gx = γ·rstd·(gy − (x̂·Σgy·x̂ + Σgy)/M) elementwise
//
// Both NCHW-contiguous and channels-last layouts are native: channels-last
// (the MIOpen-friendly conv layout) reduces columns of an [M, C] matrix —
// lanes own channels, accesses stay 16 B-coalesced with zero bank traffic.

#include <torch/extension.h>
#include <ATen/Dispatch.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

// ---------- NCHW: x[n][c][s], reduce over n and s ------------------------
// grid.x = C, grid.y = blocks per channel; threads stride over (n, s).
template <typename T, bool BWD>
__global__ void bn_reduce_nchw_kernel(const T* __restrict__ x,
                                      const T* __restrict__ gy,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ rstd,
                                      float* __restrict__ out0,  // Σx | Σgy
                                      float* __restrict__ out1,  // Σx²| Σgy·x̂
                                      int C, long N, long S) {
  __shared__ float scratch[2][CMN_BLOCK / CMN_WAVE];
  const int c = blockIdx.x;
  const long per = N * S;
  const long stride = (long)gridDim.y * blockDim.x;
  float a0 = 0.f, a1 = 0.f;
  const float mu = BWD ? mean[c] : 0.f;
  const float rs = BWD ? rstd[c] : 0.f;
  for (long i = blockIdx.y * (long)blockDim.x + threadIdx.x; i < per;
       i += stride) {
    const long n = i / S;
    const long s = i - n * S;
    const long idx = (n * C + c) * S + s;
    if (BWD) {
      const float g = float(gy[idx]);
      const float xh = (float(x[idx]) - mu) * rs;
      a0 += g;
      a1 += g * xh;
    } else {
      const float v = float(x[idx]);
      a0 += v;
      a1 += v * v;
    }
  }
  const int lane = threadIdx.x & (CMN_WAVE - 1);
  const int wid = threadIdx.x / CMN_WAVE;
  a0 = wave_reduce_sum(a0);
  a1 = wave_reduce_sum(a1);
  if (lane == 0) { scratch[0][wid] = a0; scratch[1][wid] = a1; }
  __syncthreads();
  if (wid == 0) {
    const int nw = blockDim.x / CMN_WAVE;
    float r0 = (lane < nw) ? scratch[0][lane] : 0.f;
    float r1 = (lane < nw) ? scratch[1][lane] : 0.f;
    r0 = wave_reduce_sum(r0);
    r1 = wave_reduce_sum(r1);
    if (lane == 0) {
      atomicAdd(&out0[c], r0);
      atomicAdd(&out1[c], r1);
    }
  }
}

// ---------- channels-last: x[m][c] (m = N*S), reduce over m --------------
// Thread t owns channels [4*(t%ct), 4*(t%ct)+4) within a channel tile;
// rows are covered by (blockIdx, t/ct).  C % 4 == 0 path.
template <typename T, bool BWD>
__global__ void bn_reduce_cl_kernel(const T* __restrict__ x,
                                    const T* __restrict__ gy,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ out0,
                                    float* __restrict__ out1,
                                    int C, long M) {
  // channel tile of 1024 (=256 threads × 4); loop tiles when C > 1024.
  const int ct = min(C / 4, (int)blockDim.x);          // threads per row
  const int rows_per_block = blockDim.x / ct;
  const int tile_ch = ct * 4;
  for (int c_base = 0; c_base < C; c_base += tile_ch) {
    const int tc = threadIdx.x % ct;
    const int tr = threadIdx.x / ct;
    const int c = c_base + tc * 4;
    float a[8] = {0, 0, 0, 0, 0, 0, 0, 0};  // 4×(sum, sq)
    if (c + 4 <= C && tr < rows_per_block) {
      float m4[4] = {0, 0, 0, 0}, r4[4] = {0, 0, 0, 0};
      if (BWD) {
#pragma unroll
        for (int k = 0; k < 4; ++k) { m4[k] = mean[c + k]; r4[k] = rstd[c + k]; }
      }
      const long row_stride = (long)gridDim.x * rows_per_block;
      for (long m = blockIdx.x * (long)rows_per_block + tr; m < M;
           m += row_stride) {
        const long idx = m * C + c;
        if (BWD) {
          const uint2 vg = *reinterpret_cast<const uint2*>(gy + idx);
          const uint2 vx = *reinterpret_cast<const uint2*>(x + idx);
          const unsigned ug[2] = {vg.x, vg.y}, ux[2] = {vx.x, vx.y};
#pragma unroll
          for (int h = 0; h < 2; ++h) {
#pragma unroll
            for (int l = 0; l < 2; ++l) {
              const int k = 2 * h + l;
              const T gv = T(uint16_t(l ? (ug[h] >> 16) : (ug[h] & 0xffffu)),
                             typename T::from_bits_t());
              const T xv = T(uint16_t(l ? (ux[h] >> 16) : (ux[h] & 0xffffu)),
                             typename T::from_bits_t());
              const float g = float(gv);
              a[2 * k] += g;
              a[2 * k + 1] += g * ((float(xv) - m4[k]) * r4[k]);
            }
          }
        } else {
          const uint2 vx = *reinterpret_cast<const uint2*>(x + idx);
          const unsigned ux[2] = {vx.x, vx.y};
#pragma unroll
          for (int h = 0; h < 2; ++h) {
#pragma unroll
            for (int l = 0; l < 2; ++l) {
              const int k = 2 * h + l;
              const T xv = T(uint16_t(l ? (ux[h] >> 16) : (ux[h] & 0xffffu)),
                             typename T::from_bits_t());
              const float v = float(xv);
              a[2 * k] += v;
              a[2 * k + 1] += v * v;
            }
          }
        }
      }
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        if (a[2 * k] != 0.f || a[2 * k + 1] != 0.f) {
          atomicAdd(&out0[c + k], a[2 * k]);
          atomicAdd(&out1[c + k], a[2 * k + 1]);
        }
      }
    }
  }
}

// fp32 channels-last falls back to a scalar-indexed variant.
template <typename T, bool BWD>
__global__ void bn_reduce_cl_scalar_kernel(const T* __restrict__ x,
                                           const T* __restrict__ gy,
                                           const float* __restrict__ mean,
                                           const float* __restrict__ rstd,
                                           float* __restrict__ out0,
                                           float* __restrict__ out1,
                                           int C, long M) {
  const int c = blockIdx.x % C;
  const int blk = blockIdx.x / C;
  const int nblk = gridDim.x / C;
  if (blk >= nblk) return;
  __shared__ float scratch[2][CMN_BLOCK / CMN_WAVE];
  float a0 = 0.f, a1 = 0.f;
  const float mu = BWD ? mean[c] : 0.f;
  const float rs = BWD ? rstd[c] : 0.f;
  const long stride = (long)nblk * blockDim.x;
  for (long m = blk * (long)blockDim.x + threadIdx.x; m < M; m += stride) {
    const long idx = m * C + c;
    if (BWD) {
      const float g = float(gy[idx]);
      a0 += g;
      a1 += g * ((float(x[idx]) - mu) * rs);
    } else {
      const float v = float(x[idx]);
      a0 += v;
      a1 += v * v;
    }
  }
  const int lane = threadIdx.x & (CMN_WAVE - 1);
  const int wid = threadIdx.x / CMN_WAVE;
  a0 = wave_reduce_sum(a0);
  a1 = wave_reduce_sum(a1);
  if (lane == 0) { scratch[0][wid] = a0; scratch[1][wid] = a1; }
  __syncthreads();
  if (wid == 0) {
    const int nw = blockDim.x / CMN_WAVE;
    float r0 = (lane < nw) ? scratch[0][lane] : 0.f;
    float r1 = (lane < nw) ? scratch[1][lane] : 0.f;
    r0 = wave_reduce_sum(r0);
    r1 = wave_reduce_sum(r1);
    if (lane == 0) { atomicAdd(&out0[c], r0); atomicAdd(&out1[c], r1); }
  }
}

// ---------- elementwise normalize / input-grad ---------------------------
// MODE 0: y = (x-μ)·rstd·γ + β
// MODE 1: gx = γ·rstd·(gy − (x̂·sgx + sg)/M)
template <typename T, int MODE, bool CHANNELS_LAST>
__global__ void bn_pointwise_kernel(const T* __restrict__ x,
                                    const T* __restrict__ gy,
                                    T* __restrict__ out,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ beta,
                                    const float* __restrict__ sg,   // Σgy
                                    const float* __restrict__ sgx,  // Σgy·x̂
                                    float inv_m, int C, long S, long total) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int c = CHANNELS_LAST ? (int)(i % C) : (int)((i / S) % C);
    const float xv = float(x[i]);
    if (MODE == 0) {
      const float xh = (xv - mean[c]) * rstd[c];
      out[i] = T(xh * gamma[c] + beta[c]);
    } else {
      const float xh = (xv - mean[c]) * rstd[c];
      const float g = float(gy[i]);
      out[i] = T(gamma[c] * rstd[c] * (g - (xh * sgx[c] + sg[c]) * inv_m));
    }
  }
}

bool is_channels_last(const torch::Tensor& x) {
  return x.dim() == 4 &&
         x.is_contiguous(at::MemoryFormat::ChannelsLast) &&
         !x.is_contiguous();
}

}  // namespace

#define DISPATCH_X(TENSOR, ...)                                              \
  CMN_DISPATCH(TENSOR.scalar_type(), "bn_x", [&] {                           \
    return __VA_ARGS__();                                                    \
  })

static void bn_reduce(const torch::Tensor& x, const torch::Tensor* gy,
                      const torch::Tensor* mean, const torch::Tensor* rstd,
                      torch::Tensor& out0, torch::Tensor& out1) {
  const int C = (int)x.size(1);
  const long total = x.numel();
  const long M = total / C;
  auto stream = c10::hip::getCurrentHIPStream();
  out0.zero_();
  out1.zero_();
  const bool cl = is_channels_last(x);
  const bool bwd = gy != nullptr;
  DISPATCH_X(x, [&] {
    const scalar_t* xp = (const scalar_t*)x.data_ptr();
    const scalar_t* gp = bwd ? (const scalar_t*)gy->data_ptr() : nullptr;
    const float* mp = bwd ? mean->data_ptr<float>() : nullptr;
    const float* rp = bwd ? rstd->data_ptr<float>() : nullptr;
    auto run = [&](auto bwd_tag) {
      constexpr bool IS_BWD = decltype(bwd_tag)::value;
      if (cl) {
        constexpr bool is16 = std::is_same<scalar_t, c10::BFloat16>::value ||
                              std::is_same<scalar_t, c10::Half>::value;
        if (is16 && C % 4 == 0) {
          if constexpr (is16) {
            const int ct = std::min<long>(C / 4, CMN_BLOCK);
            const int rpb = CMN_BLOCK / ct;
            const int grid = (int)std::min<long>((M + rpb - 1) / rpb, 1024);
            hipLaunchKernelGGL((bn_reduce_cl_kernel<scalar_t, IS_BWD>),
                               dim3(grid), dim3(CMN_BLOCK), 0,
                               stream.stream(), xp, gp, mp, rp,
                               out0.data_ptr<float>(),
                               out1.data_ptr<float>(), C, M);
          }
        } else {
          const int nblk = (int)std::min<long>(
              std::max<long>(2048 / C, 1L),
              (M + CMN_BLOCK - 1) / CMN_BLOCK);
          hipLaunchKernelGGL((bn_reduce_cl_scalar_kernel<scalar_t, IS_BWD>),
                             dim3(C * nblk), dim3(CMN_BLOCK), 0,
                             stream.stream(), xp, gp, mp, rp,
                             out0.data_ptr<float>(), out1.data_ptr<float>(),
                             C, M);
        }
      } else {
        const long N = x.size(0);
        const long S = M / N;
        const int per_c =
            (int)std::min<long>((N * S + CMN_BLOCK * 4 - 1) / (CMN_BLOCK * 4),
                                std::max<long>(2048 / C, 1L));
        hipLaunchKernelGGL((bn_reduce_nchw_kernel<scalar_t, IS_BWD>),
                           dim3(C, std::max(per_c, 1)), dim3(CMN_BLOCK), 0,
                           stream.stream(), xp, gp, mp, rp,
                           out0.data_ptr<float>(), out1.data_ptr<float>(),
                           C, N, S);
      }
    };
    if (bwd) run(std::true_type{});
    else run(std::false_type{});
  });
}

std::tuple<torch::Tensor, torch::Tensor, long> bn_stats(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() >= 2, "bn_stats: GPU, >=2-D");
  const int C = (int)x.size(1);
  auto opts = torch::TensorOptions().dtype(at::kFloat).device(x.device());
  auto s = torch::empty({C}, opts);
  auto sq = torch::empty({C}, opts);
  bn_reduce(x, nullptr, nullptr, nullptr, s, sq);
  return {s, sq, x.numel() / C};
}

std::tuple<torch::Tensor, torch::Tensor> bn_bwd_reduce(torch::Tensor x,
                                                       torch::Tensor gy,
                                                       torch::Tensor mean,
                                                       torch::Tensor rstd) {
  TORCH_CHECK(x.is_cuda() && gy.sizes() == x.sizes());
  const int C = (int)x.size(1);
  auto opts = torch::TensorOptions().dtype(at::kFloat).device(x.device());
  auto gbeta = torch::empty({C}, opts);
  auto ggamma = torch::empty({C}, opts);
  bn_reduce(x, &gy, &mean, &rstd, gbeta, ggamma);
  return {gbeta, ggamma};
}

torch::Tensor bn_fwd(torch::Tensor x, torch::Tensor mean, torch::Tensor rstd,
                     torch::Tensor gamma, torch::Tensor beta) {
  TORCH_CHECK(x.is_cuda());
  const bool cl = is_channels_last(x);
  auto y = torch::empty_like(x);
  const int C = (int)x.size(1);
  const long total = x.numel();
  const long S = (x.dim() > 2 && !cl) ? x.numel() / (x.size(0) * C) : 1;
  auto stream = c10::hip::getCurrentHIPStream();
  DISPATCH_X(x, [&] {
    auto launch = [&](auto cl_tag) {
      hipLaunchKernelGGL(
          (bn_pointwise_kernel<scalar_t, 0, decltype(cl_tag)::value>),
          dim3(cmn_grid(total, CMN_BLOCK)), dim3(CMN_BLOCK), 0,
          stream.stream(), (const scalar_t*)x.data_ptr(), nullptr,
          (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
          rstd.data_ptr<float>(), gamma.data_ptr<float>(),
          beta.data_ptr<float>(), nullptr, nullptr, 0.f, C, S, total);
    };
    if (cl) launch(std::true_type{});
    else launch(std::false_type{});
  });
  return y;
}

torch::Tensor bn_bwd_input(torch::Tensor x, torch::Tensor gy,
                           torch::Tensor mean, torch::Tensor rstd,
                           torch::Tensor gamma, torch::Tensor gbeta,
                           torch::Tensor ggamma, double total_count) {
  TORCH_CHECK(x.is_cuda());
  const bool cl = is_channels_last(x);
  auto gx = torch::empty_like(x);
  const int C = (int)x.size(1);
  const long total = x.numel();
  const long S = (x.dim() > 2 && !cl) ? x.numel() / (x.size(0) * C) : 1;
  auto stream = c10::hip::getCurrentHIPStream();
  DISPATCH_X(x, [&] {
    auto launch = [&](auto cl_tag) {
      hipLaunchKernelGGL(
          (bn_pointwise_kernel<scalar_t, 1, decltype(cl_tag)::value>),
          dim3(cmn_grid(total, CMN_BLOCK)), dim3(CMN_BLOCK), 0,
          stream.stream(), (const scalar_t*)x.data_ptr(),
          (const scalar_t*)gy.data_ptr(), (scalar_t*)gx.data_ptr(),
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gamma.data_ptr<float>(), nullptr, gbeta.data_ptr<float>(),
          ggamma.data_ptr<float>(), (float)(1.0 / total_count), C, S,
          total);
    };
    if (cl) launch(std::true_type{});
    else launch(std::false_type{});
  });
  return gx;
}
