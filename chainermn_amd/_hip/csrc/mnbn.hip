// BatchNorm device kernels (gfx950) — used by MultiNodeBatchNormalization
// (cross-rank global-batch BN), the local
// FusedBatchNorm2d module, and the BN+ReLU / BN+Add+ReLU fusions that
// replace torch's separate elementwise kernels in the ResNet hot path.
//
//   bn_stats      : per-channel (Σx, Σx²) fused pass (2×C+1 buffer → RCCL)
//   bn_finalize   : mean/rstd/var (+ fused running-stat update), one launch
//   bn_fwd        : y = act(bn(x) [+ res]);  act ∈ {id, relu}
//   bn_bwd_reduce : per-channel (Σgy', Σgy'·x̂),  gy' = gy·relu-mask
//   bn_bwd_input  : gx = γ·rstd·(gy' − (x̂·Σgy'x̂ + Σgy')/M) [+ gres out]
//
// relu-mask modes (RM): 0 = none; 1 = AFFINE (mask = γ·x̂+β > 0 — free,
// recomputed from x, no extra tensor); 2 = FROM_Y (mask = y > 0 — needed
// when a residual was added before the relu; reads the saved output).
//
// Perf notes (vs torch native channels_last BN at 66% of a ResNet-50 step,
// see profiles/): channels-last reduce = 8 channels/lane uint4 loads,
// per-thread fp32 accumulators, LDS cross-row tree, per-block partials +
// parallel stage-2 (same-address atomic chains and serial stage-2 each
// cost >2x); pointwise = 8 elems/lane with float4 parameter loads and one
// channel computation per vector.

#include <torch/extension.h>
#include <ATen/Dispatch.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

// ---- vector helpers ------------------------------------------------------
__device__ __forceinline__ void load4(const float* p, float (&r)[4]) {
  const float4 v = *reinterpret_cast<const float4*>(p);
  r[0] = v.x; r[1] = v.y; r[2] = v.z; r[3] = v.w;
}
template <typename T16>
__device__ __forceinline__ void load4(const T16* p, float (&r)[4]) {
  const uint2 v = *reinterpret_cast<const uint2*>(p);
  const unsigned u[2] = {v.x, v.y};
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    r[2 * h] = float(T16(uint16_t(u[h] & 0xffffu),
                         typename T16::from_bits_t()));
    r[2 * h + 1] = float(T16(uint16_t(u[h] >> 16),
                             typename T16::from_bits_t()));
  }
}
__device__ __forceinline__ void loadp8(const float* p, int c0,
                                       float (&r)[8]) {
  const float4 a = *reinterpret_cast<const float4*>(p + c0);
  const float4 b = *reinterpret_cast<const float4*>(p + c0 + 4);
  r[0] = a.x; r[1] = a.y; r[2] = a.z; r[3] = a.w;
  r[4] = b.x; r[5] = b.y; r[6] = b.z; r[7] = b.w;
}
template <typename T16>
__device__ __forceinline__ void loadp8(const T16* p, int c0,
                                       float (&r)[8]) {
  load4(p + c0, *(float(*)[4])(r));
  load4(p + c0 + 4, *(float(*)[4])(r + 4));
}

// ---- NCHW reduce (scalar fallback; bench path is channels-last) ----------
template <typename T, bool BWD, int RM>
__global__ void bn_reduce_nchw_kernel(const T* __restrict__ x,
                                      const T* __restrict__ gy,
                                      const T* __restrict__ yout,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ rstd,
                                      const T* __restrict__ gamma,
                                      const T* __restrict__ beta,
                                      float* __restrict__ out0,
                                      float* __restrict__ out1,
                                      int C, long N, long S) {
  __shared__ float scratch[2][CMN_BLOCK / CMN_WAVE];
  const int c = blockIdx.x;
  const long per = N * S;
  const long stride = (long)gridDim.y * blockDim.x;
  float a0 = 0.f, a1 = 0.f;
  const float mu = BWD ? mean[c] : 0.f;
  const float rs = BWD ? rstd[c] : 0.f;
  const float gm = (BWD && RM == 1) ? float(gamma[c]) : 0.f;
  const float bt = (BWD && RM == 1) ? float(beta[c]) : 0.f;
  for (long i = blockIdx.y * (long)blockDim.x + threadIdx.x; i < per;
       i += stride) {
    const long n = i / S;
    const long s = i - n * S;
    const long idx = (n * C + c) * S + s;
    if (BWD) {
      float g = float(gy[idx]);
      const float xh = (float(x[idx]) - mu) * rs;
      if (RM == 1) g = (xh * gm + bt) > 0.f ? g : 0.f;
      if (RM == 2) g = float(yout[idx]) > 0.f ? g : 0.f;
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

// ---- channels-last reduce: x[m][c], per-block partials -------------------
template <typename T, bool BWD, int CPT, int RM>
__global__ void bn_reduce_cl_kernel(const T* __restrict__ x,
                                    const T* __restrict__ gy,
                                    const T* __restrict__ yout,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    const T* __restrict__ gamma,
                                    const T* __restrict__ beta,
                                    float* __restrict__ partials,
                                    int C, long M) {
  __shared__ float lds[CMN_BLOCK * 2 * CPT > 16384 ? 16384
                                                   : CMN_BLOCK * 2 * CPT];
  const int ct = min(C / CPT, (int)blockDim.x);
  const int rpb = blockDim.x / ct;
  const int tile_ch = ct * CPT;
  const int tc = threadIdx.x % ct;
  const int tr = threadIdx.x / ct;
  for (int c_base = 0; c_base < C; c_base += tile_ch) {
    const int c = c_base + tc * CPT;
    // Last tile when C > ct*CPT: threads whose channel slice starts past C
    // must not touch memory (the caller guarantees C % CPT == 0, so
    // c < C implies the whole [c, c+CPT) slice is in bounds). They still
    // write their zeroed accumulator into LDS for the cross-row tree.
    const bool in_c = c < C;
    float a[2 * CPT];
#pragma unroll
    for (int k = 0; k < 2 * CPT; ++k) a[k] = 0.f;
    float m4[CPT], r4[CPT], g4[CPT], b4[CPT];
    if (BWD && in_c) {
#pragma unroll
      for (int k = 0; k < CPT; ++k) {
        m4[k] = mean[c + k];
        r4[k] = rstd[c + k];
        if (RM == 1) {
          g4[k] = float(gamma[c + k]);
          b4[k] = float(beta[c + k]);
        }
      }
    }
    const long row_stride = (long)gridDim.x * rpb;
    if (tr < rpb && in_c) {
      for (long m = blockIdx.x * (long)rpb + tr; m < M; m += row_stride) {
        const long idx = m * C + c;
        float xv[CPT], gv[CPT], yv[CPT];
#pragma unroll
        for (int q = 0; q < CPT / 4; ++q) {
          load4(x + idx + 4 * q, *(float(*)[4])(xv + 4 * q));
          if (BWD) load4(gy + idx + 4 * q, *(float(*)[4])(gv + 4 * q));
          if (BWD && RM == 2)
            load4(yout + idx + 4 * q, *(float(*)[4])(yv + 4 * q));
        }
#pragma unroll
        for (int k = 0; k < CPT; ++k) {
          if (BWD) {
            const float xh = (xv[k] - m4[k]) * r4[k];
            float g = gv[k];
            if (RM == 1) g = (xh * g4[k] + b4[k]) > 0.f ? g : 0.f;
            if (RM == 2) g = yv[k] > 0.f ? g : 0.f;
            a[2 * k] += g;
            a[2 * k + 1] += g * xh;
          } else {
            a[2 * k] += xv[k];
            a[2 * k + 1] += xv[k] * xv[k];
          }
        }
      }
    }
#pragma unroll
    for (int k = 0; k < 2 * CPT; ++k)
      lds[threadIdx.x * 2 * CPT + k] = a[k];
    __syncthreads();
    // cross-row tree (non-power-of-two-safe ceil halving)
    for (int rem = rpb; rem > 1;) {
      const int off = (rem + 1) >> 1;
      if (tr < rem - off) {
#pragma unroll
        for (int k = 0; k < 2 * CPT; ++k)
          lds[threadIdx.x * 2 * CPT + k] +=
              lds[(threadIdx.x + off * ct) * 2 * CPT + k];
      }
      __syncthreads();
      rem = off;
    }
    if (tr == 0 && in_c) {
      // per-block partials: no same-address atomic chains
#pragma unroll
      for (int k = 0; k < CPT; ++k) {
        partials[blockIdx.x * 2 * C + c + k] =
            lds[threadIdx.x * 2 * CPT + 2 * k];
        partials[blockIdx.x * 2 * C + C + c + k] =
            lds[threadIdx.x * 2 * CPT + 2 * k + 1];
      }
    }
    __syncthreads();
  }
}

// stage 2 (hierarchical, no atomics, no pre-zeroed output — a serial
// single block cost 235 us; same-address atomics serialize).
// 2a: grid (jblocks, GS) — block (jb, gi) sums its g-slice -> mid[gi][2C]
// 2b: grid (jblocks) — 4 g-lanes × 64 j-lanes + LDS tree over GS=16 mids
#define BN_GS 16
__global__ void bn_reduce_stage2a_kernel(const float* __restrict__ partials,
                                         float* __restrict__ mid,
                                         int C, int n_blocks) {
  __shared__ float lds[CMN_BLOCK];
  const int tj = threadIdx.x % 64;
  const int tg = threadIdx.x / 64;
  const int j = blockIdx.x * 64 + tj;
  const int gi = blockIdx.y;
  float acc = 0.f;
  if (j < 2 * C) {
    for (int g = gi * 4 + tg; g < n_blocks; g += BN_GS * 4)
      acc += partials[(long)g * 2 * C + j];
  }
  lds[threadIdx.x] = acc;
  __syncthreads();
  if (tg == 0 && j < 2 * C)
    mid[(long)gi * 2 * C + j] =
        lds[tj] + lds[tj + 64] + lds[tj + 128] + lds[tj + 192];
}

__global__ void bn_reduce_stage2b_kernel(const float* __restrict__ mid,
                                         float* __restrict__ out0,
                                         float* __restrict__ out1, int C) {
  __shared__ float lds[CMN_BLOCK];
  const int tj = threadIdx.x % 64;
  const int tg = threadIdx.x / 64;
  const int j = blockIdx.x * 64 + tj;
  float acc = 0.f;
  if (j < 2 * C) {
#pragma unroll
    for (int g = tg; g < BN_GS; g += 4)
      acc += mid[(long)g * 2 * C + j];
  }
  lds[threadIdx.x] = acc;
  __syncthreads();
  if (tg == 0 && j < 2 * C) {
    acc = lds[tj] + lds[tj + 64] + lds[tj + 128] + lds[tj + 192];
    if (j < C) out0[j] = acc;
    else out1[j - C] = acc;
  }
}

// ---- pointwise -----------------------------------------------------------
// MODE 0 fwd: y = act(bn(x) + res?)    (RES: add residual; RM==1: relu)
// MODE 1 bwd: gy' = gy·mask; gx = γ·rstd·(gy' − (x̂·sgx + sg)/M);
//             gres = gy' when RES
template <typename T, int MODE, bool CHANNELS_LAST, int RM, bool RES>
__global__ void bn_pointwise_vec_kernel(const T* __restrict__ x,
                                        const T* __restrict__ gy,
                                        const T* __restrict__ aux,  // res|y
                                        T* __restrict__ out,
                                        T* __restrict__ gres,
                                        const float* __restrict__ mean,
                                        const float* __restrict__ rstd,
                                        const T* __restrict__ gamma,
                                        const T* __restrict__ beta,
                                        const float* __restrict__ sg,
                                        const float* __restrict__ sgx,
                                        float inv_m, int C, long S,
                                        long total8) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = blockIdx.x * (long)blockDim.x + threadIdx.x; v < total8;
       v += stride) {
    const long base = v * 8;
    float xv[8], gv[8], av[8], o[8], gr[8];
    V8<T>::load(x + base, xv);
    if (MODE == 1) V8<T>::load(gy + base, gv);
    if ((MODE == 0 && RES) || (MODE == 1 && RM == 2))
      V8<T>::load(aux + base, av);
    float mn[8], rs[8], gm[8], bt[8], s0[8], s1[8];
    if (CHANNELS_LAST) {
      const int c0 = (int)(base % (long)C);
      loadp8(mean, c0, mn);
      loadp8(rstd, c0, rs);
      loadp8(gamma, c0, gm);
      if (MODE == 0 || RM == 1) loadp8(beta, c0, bt);
      if (MODE == 1) {
        loadp8(sg, c0, s0);
        loadp8(sgx, c0, s1);
      }
    } else {
      const int c = (int)((base / S) % C);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        mn[k] = mean[c]; rs[k] = rstd[c]; gm[k] = gamma[c];
        if (MODE == 0 || RM == 1) bt[k] = beta[c];
        if (MODE == 1) { s0[k] = sg[c]; s1[k] = sgx[c]; }
      }
    }
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      const float xh = (xv[k] - mn[k]) * rs[k];
      if (MODE == 0) {
        float y = xh * gm[k] + bt[k];
        if (RES) y += av[k];
        if (RM == 1) y = fmaxf(y, 0.f);
        o[k] = y;
      } else {
        float g = gv[k];
        if (RM == 1) g = (xh * gm[k] + bt[k]) > 0.f ? g : 0.f;
        if (RM == 2) g = av[k] > 0.f ? g : 0.f;
        if (RES) gr[k] = g;
        o[k] = gm[k] * rs[k] * (g - (xh * s1[k] + s0[k]) * inv_m);
      }
    }
    V8<T>::store(out + base, o);
    if (MODE == 1 && RES) V8<T>::store(gres + base, gr);
  }
}

template <typename T, int MODE, bool CHANNELS_LAST, int RM, bool RES>
__global__ void bn_pointwise_kernel(const T* __restrict__ x,
                                    const T* __restrict__ gy,
                                    const T* __restrict__ aux,
                                    T* __restrict__ out,
                                    T* __restrict__ gres,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    const T* __restrict__ gamma,
                                    const T* __restrict__ beta,
                                    const float* __restrict__ sg,
                                    const float* __restrict__ sgx,
                                    float inv_m, int C, long S, long total) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int c = CHANNELS_LAST ? (int)(i % C) : (int)((i / S) % C);
    const float xh = (float(x[i]) - mean[c]) * rstd[c];
    if (MODE == 0) {
      float y = xh * float(gamma[c]) + float(beta[c]);
      if (RES) y += float(aux[i]);
      if (RM == 1) y = fmaxf(y, 0.f);
      out[i] = T(y);
    } else {
      float g = float(gy[i]);
      if (RM == 1) g = (xh * float(gamma[c]) + float(beta[c])) > 0.f
                        ? g : 0.f;
      if (RM == 2) g = float(aux[i]) > 0.f ? g : 0.f;
      if (RES) gres[i] = T(g);
      out[i] = T(float(gamma[c]) * rstd[c]
                 * (g - (xh * sgx[c] + sg[c]) * inv_m));
    }
  }
}

// ---- finalize ------------------------------------------------------------
__global__ void bn_finalize_kernel(const float* __restrict__ buf,
                                   float* __restrict__ mean,
                                   float* __restrict__ rstd,
                                   float* __restrict__ var,
                                   float* __restrict__ r_mean,
                                   float* __restrict__ r_var,
                                   float momentum, float unbias,
                                   float count, float eps, int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  const float inv = 1.f / (count > 0.f ? count : buf[2 * C]);
  if (c < C) {
    const float mu = buf[c] * inv;
    const float v = fmaxf(buf[C + c] * inv - mu * mu, 0.f);
    mean[c] = mu;
    var[c] = v;
    rstd[c] = rsqrtf(v + eps);
    if (r_mean != nullptr) {
      r_mean[c] = (1.f - momentum) * r_mean[c] + momentum * mu;
      r_var[c] = (1.f - momentum) * r_var[c] + momentum * v * unbias;
    }
  }
}

bool is_channels_last(const torch::Tensor& x) {
  return x.dim() == 4 &&
         x.is_contiguous(at::MemoryFormat::ChannelsLast) &&
         !x.is_contiguous();
}

template <typename F>
void static_rm(int rm, F f) {
  using i0 = std::integral_constant<int, 0>;
  using i1 = std::integral_constant<int, 1>;
  using i2 = std::integral_constant<int, 2>;
  if (rm == 0) f(i0{});
  else if (rm == 1) f(i1{});
  else f(i2{});
}

}  // namespace

#define DISPATCH_X(TENSOR, ...)                                              \
  CMN_DISPATCH(TENSOR.scalar_type(), "bn_x", [&] {                           \
    return __VA_ARGS__();                                                    \
  })

static void bn_reduce(const torch::Tensor& x, const torch::Tensor* gy,
                      const torch::Tensor* yout,
                      const torch::Tensor* mean, const torch::Tensor* rstd,
                      const torch::Tensor* gamma, const torch::Tensor* beta,
                      int rm, torch::Tensor& out0, torch::Tensor& out1) {
  const int C = (int)x.size(1);
  const long total = x.numel();
  const long M = total / C;
  auto stream = c10::hip::getCurrentHIPStream();
  const bool cl = is_channels_last(x) || x.dim() == 2;
  const bool bwd = gy != nullptr;
  DISPATCH_X(x, [&] {
    const scalar_t* xp = (const scalar_t*)x.data_ptr();
    const scalar_t* gp = bwd ? (const scalar_t*)gy->data_ptr() : nullptr;
    const scalar_t* yp = yout ? (const scalar_t*)yout->data_ptr() : nullptr;
    const float* mp = bwd ? mean->data_ptr<float>() : nullptr;
    const float* rp = bwd ? rstd->data_ptr<float>() : nullptr;
    const scalar_t* gmp = gamma
        ? (const scalar_t*)gamma->data_ptr() : nullptr;
    const scalar_t* btp = beta
        ? (const scalar_t*)beta->data_ptr() : nullptr;
    auto run = [&](auto bwd_tag, auto rm_tag) {
      constexpr bool IS_BWD = decltype(bwd_tag)::value;
      constexpr int RM = decltype(rm_tag)::value;
      if (cl && C % 4 == 0) {
        const int cpt = (C % 8 == 0) ? 8 : 4;
        const int ct = std::min<long>(C / cpt, CMN_BLOCK);
        const int rpb = CMN_BLOCK / ct;
        const int grid = (int)std::min<long>((M + rpb - 1) / rpb, 1024);
        auto partials = torch::empty({grid, 2L * C},
            torch::TensorOptions().dtype(at::kFloat).device(x.device()));
        float* pp = partials.data_ptr<float>();
        if (cpt == 8) {
          hipLaunchKernelGGL((bn_reduce_cl_kernel<scalar_t, IS_BWD, 8, RM>),
                             dim3(grid), dim3(CMN_BLOCK), 0,
                             stream.stream(), xp, gp, yp, mp, rp, gmp, btp,
                             pp, C, M);
        } else {
          hipLaunchKernelGGL((bn_reduce_cl_kernel<scalar_t, IS_BWD, 4, RM>),
                             dim3(grid), dim3(CMN_BLOCK), 0,
                             stream.stream(), xp, gp, yp, mp, rp, gmp, btp,
                             pp, C, M);
        }
        auto mid = torch::empty({BN_GS, 2L * C},
            torch::TensorOptions().dtype(at::kFloat).device(x.device()));
        const int jblocks = (2 * C + 63) / 64;
        hipLaunchKernelGGL(bn_reduce_stage2a_kernel,
                           dim3(jblocks, BN_GS), dim3(CMN_BLOCK), 0,
                           stream.stream(), pp, mid.data_ptr<float>(), C,
                           grid);
        hipLaunchKernelGGL(bn_reduce_stage2b_kernel, dim3(jblocks),
                           dim3(CMN_BLOCK), 0, stream.stream(),
                           mid.data_ptr<float>(), out0.data_ptr<float>(),
                           out1.data_ptr<float>(), C);
      } else {
        out0.zero_();
        out1.zero_();
        long N = x.size(0), S;
        if (cl) {  // [M, C] with C % 4 != 0: per-channel scalar blocks
          hipLaunchKernelGGL((bn_reduce_nchw_kernel<scalar_t, IS_BWD, RM>),
                             dim3(C, 8), dim3(CMN_BLOCK), 0,
                             stream.stream(), xp, gp, yp, mp, rp, gmp, btp,
                             out0.data_ptr<float>(), out1.data_ptr<float>(),
                             C, M, 1L);
        } else {
          S = M / N;
          const int per_c = (int)std::min<long>(
              (N * S + CMN_BLOCK * 4 - 1) / (CMN_BLOCK * 4),
              std::max<long>(2048 / C, 1L));
          hipLaunchKernelGGL((bn_reduce_nchw_kernel<scalar_t, IS_BWD, RM>),
                             dim3(C, std::max(per_c, 1)), dim3(CMN_BLOCK),
                             0, stream.stream(), xp, gp, yp, mp, rp, gmp,
                             btp, out0.data_ptr<float>(),
                             out1.data_ptr<float>(), C, N, S);
        }
      }
    };
    static_rm(bwd ? rm : 0, [&](auto rm_tag) {
      if (bwd) run(std::true_type{}, rm_tag);
      else run(std::false_type{}, rm_tag);
    });
  });
}

static void bn_pointwise(int mode, const torch::Tensor& x,
                         const torch::Tensor* gy, const torch::Tensor* aux,
                         torch::Tensor& out, torch::Tensor* gres, int rm,
                         const torch::Tensor& mean, const torch::Tensor& rstd,
                         const torch::Tensor& gamma,
                         const torch::Tensor* beta, const torch::Tensor* sg,
                         const torch::Tensor* sgx, float inv_m) {
  const bool cl = is_channels_last(x) || x.dim() == 2;
  const int C = (int)x.size(1);
  const long total = x.numel();
  const long S = (x.dim() > 2 && !cl) ? total / (x.size(0) * C) : 1;
  auto stream = c10::hip::getCurrentHIPStream();
  const bool vec = cl ? (C % 8 == 0) : (S % 8 == 0);
  const bool res = gres != nullptr || (mode == 0 && aux != nullptr);
  DISPATCH_X(x, [&] {
    const scalar_t* xp = (const scalar_t*)x.data_ptr();
    const scalar_t* gp = gy ? (const scalar_t*)gy->data_ptr() : nullptr;
    const scalar_t* ap = aux ? (const scalar_t*)aux->data_ptr() : nullptr;
    scalar_t* op = (scalar_t*)out.data_ptr();
    scalar_t* grp = gres ? (scalar_t*)gres->data_ptr() : nullptr;
    const scalar_t* btp = beta
        ? (const scalar_t*)beta->data_ptr() : nullptr;
    const float* sgp = sg ? sg->data_ptr<float>() : nullptr;
    const float* sgxp = sgx ? sgx->data_ptr<float>() : nullptr;
    auto launch = [&](auto mode_tag, auto cl_tag, auto rm_tag,
                      auto res_tag) {
      constexpr int MODE = decltype(mode_tag)::value;
      constexpr bool CL = decltype(cl_tag)::value;
      constexpr int RM = decltype(rm_tag)::value;
      constexpr bool RES = decltype(res_tag)::value;
      if (vec) {
        hipLaunchKernelGGL(
            (bn_pointwise_vec_kernel<scalar_t, MODE, CL, RM, RES>),
            dim3(cmn_grid(total / 8, CMN_BLOCK)), dim3(CMN_BLOCK), 0,
            stream.stream(), xp, gp, ap, op, grp, mean.data_ptr<float>(),
            rstd.data_ptr<float>(), (const scalar_t*)gamma.data_ptr(),
            btp, sgp, sgxp, inv_m, C, S, total / 8);
      } else {
        hipLaunchKernelGGL(
            (bn_pointwise_kernel<scalar_t, MODE, CL, RM, RES>),
            dim3(cmn_grid(total, CMN_BLOCK)), dim3(CMN_BLOCK), 0,
            stream.stream(), xp, gp, ap, op, grp, mean.data_ptr<float>(),
            rstd.data_ptr<float>(), (const scalar_t*)gamma.data_ptr(),
            btp, sgp, sgxp, inv_m, C, S, total);
      }
    };
    using i0 = std::integral_constant<int, 0>;
    using i1 = std::integral_constant<int, 1>;
    static_rm(rm, [&](auto rm_tag) {
      auto with_mode = [&](auto mode_tag) {
        if (cl) {
          if (res) launch(mode_tag, std::true_type{}, rm_tag,
                          std::true_type{});
          else launch(mode_tag, std::true_type{}, rm_tag,
                      std::false_type{});
        } else {
          if (res) launch(mode_tag, std::false_type{}, rm_tag,
                          std::true_type{});
          else launch(mode_tag, std::false_type{}, rm_tag,
                      std::false_type{});
        }
      };
      if (mode == 0) with_mode(i0{});
      else with_mode(i1{});
    });
  });
}

// ---- public API ----------------------------------------------------------

void bn_stats_into(torch::Tensor x, torch::Tensor buf) {
  TORCH_CHECK(x.is_cuda() && buf.scalar_type() == at::kFloat);
  const int C = (int)x.size(1);
  TORCH_CHECK(buf.numel() >= 2 * C);
  auto out0 = buf.narrow(0, 0, C);
  auto out1 = buf.narrow(0, C, C);
  bn_reduce(x, nullptr, nullptr, nullptr, nullptr, nullptr, nullptr, 0,
            out0, out1);
}

void bn_bwd_reduce_act_into(torch::Tensor x, torch::Tensor gy,
                            torch::Tensor mean, torch::Tensor rstd,
                            torch::Tensor gamma, torch::Tensor beta,
                            torch::Tensor yout, long relu_mode,
                            torch::Tensor buf) {
  const int C = (int)x.size(1);
  TORCH_CHECK(buf.numel() >= 2 * C);
  auto out0 = buf.narrow(0, 0, C);
  auto out1 = buf.narrow(0, C, C);
  const torch::Tensor* yp = relu_mode == 2 ? &yout : nullptr;
  bn_reduce(x, &gy, yp, &mean, &rstd, &gamma, &beta, (int)relu_mode, out0,
            out1);
}

void bn_bwd_reduce_into(torch::Tensor x, torch::Tensor gy,
                        torch::Tensor mean, torch::Tensor rstd,
                        torch::Tensor buf) {
  bn_bwd_reduce_act_into(x, gy, mean, rstd, x, x, torch::Tensor(), 0, buf);
}

std::tuple<torch::Tensor, torch::Tensor, long> bn_stats(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() >= 2, "bn_stats: GPU, >=2-D");
  const int C = (int)x.size(1);
  auto opts = torch::TensorOptions().dtype(at::kFloat).device(x.device());
  auto s = torch::empty({C}, opts);
  auto sq = torch::empty({C}, opts);
  bn_reduce(x, nullptr, nullptr, nullptr, nullptr, nullptr, nullptr, 0, s,
            sq);
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
  bn_reduce(x, &gy, nullptr, &mean, &rstd, nullptr, nullptr, 0, gbeta,
            ggamma);
  return {gbeta, ggamma};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> bn_finalize(
    torch::Tensor buf, double count, double eps, double momentum,
    double unbias, torch::Tensor r_mean, torch::Tensor r_var) {
  const int C = (int)(count > 0 ? buf.numel() / 2 : (buf.numel() - 1) / 2);
  auto opts = torch::TensorOptions().dtype(at::kFloat).device(buf.device());
  auto mean = torch::empty({C}, opts);
  auto rstd = torch::empty({C}, opts);
  auto var = torch::empty({C}, opts);
  float* rm = nullptr;
  float* rv = nullptr;
  if (r_mean.defined() && r_mean.numel()) {
    TORCH_CHECK(r_mean.scalar_type() == at::kFloat
                && r_var.scalar_type() == at::kFloat,
                "running stats must be fp32");
    rm = r_mean.data_ptr<float>();
    rv = r_var.data_ptr<float>();
  }
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + 255) / 256), dim3(256),
                     0, stream.stream(), buf.data_ptr<float>(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     var.data_ptr<float>(), rm, rv, (float)momentum,
                     (float)unbias, (float)count, (float)eps, C);
  return {mean, rstd, var};
}

// y = act(bn(x) [+ res]); relu_mode 1 applies relu; res may be empty.
torch::Tensor bn_fwd_act(torch::Tensor x, torch::Tensor mean,
                         torch::Tensor rstd, torch::Tensor gamma,
                         torch::Tensor beta, torch::Tensor res,
                         long relu_mode) {
  TORCH_CHECK(x.is_cuda());
  auto y = torch::empty_like(x);
  const bool has_res = res.defined() && res.numel();
  const torch::Tensor* rp = has_res ? &res : nullptr;
  bn_pointwise(0, x, nullptr, rp, y, nullptr, (int)relu_mode, mean, rstd,
               gamma, &beta, nullptr, nullptr, 0.f);
  return y;
}

torch::Tensor bn_fwd(torch::Tensor x, torch::Tensor mean, torch::Tensor rstd,
                     torch::Tensor gamma, torch::Tensor beta) {
  return bn_fwd_act(x, mean, rstd, gamma, beta, torch::Tensor(), 0);
}

// gx (+ gres when want_gres); relu_mode: 0 none, 1 affine mask, 2 from y.
std::tuple<torch::Tensor, torch::Tensor> bn_bwd_input_act(
    torch::Tensor x, torch::Tensor gy, torch::Tensor mean,
    torch::Tensor rstd, torch::Tensor gamma, torch::Tensor beta,
    torch::Tensor gbeta, torch::Tensor ggamma, torch::Tensor yout,
    long relu_mode, bool want_gres, double total_count) {
  TORCH_CHECK(x.is_cuda());
  auto gx = torch::empty_like(x);
  torch::Tensor gres;
  torch::Tensor* grp = nullptr;
  if (want_gres) {
    gres = torch::empty_like(x);
    grp = &gres;
  }
  const torch::Tensor* aux = relu_mode == 2 ? &yout : nullptr;
  bn_pointwise(1, x, &gy, aux, gx, grp, (int)relu_mode, mean, rstd, gamma,
               &beta, &gbeta, &ggamma, (float)(1.0 / total_count));
  if (!want_gres) gres = torch::Tensor();
  return {gx, gres};
}

torch::Tensor bn_bwd_input(torch::Tensor x, torch::Tensor gy,
                           torch::Tensor mean, torch::Tensor rstd,
                           torch::Tensor gamma, torch::Tensor gbeta,
                           torch::Tensor ggamma, double total_count) {
  auto r = bn_bwd_input_act(x, gy, mean, rstd, gamma, gamma, gbeta, ggamma,
                            torch::Tensor(), 0, false, total_count);
  return std::get<0>(r);
}
