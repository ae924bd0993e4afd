// Fused flat-buffer optimizer kernels (gfx950).
//
// Rebuild of SURVEY.md §2.4 K5: the reference delegated per-param update
// rules to Chainer/CuPy kernels (one launch per parameter — 161 launches
// for ResNet-50).  Here the whole model updates in ONE kernel over the
// flat param/grad/state arenas, with the 1/world_size mean, the wire-dtype
// cast-back, and an optional low-precision parameter mirror all fused into
// the same HBM pass:
//
//   fused_sgd : p32 [+ p16 mirror] , v32, g in {f32,bf16,f16}
//   fused_adam: p32 [+ p16 mirror] , m32, v32, g
//   segment_sq_norms + lars_update: per-layer trust-ratio (LARS) over the
//     same arenas (segmented reduction; one workgroup covers a contiguous
//     chunk, partials land in per-segment fp32 atomics)
//
// torch.optim semantics (SGD with momentum/dampening/nesterov; Adam with
// bias correction; AdamW decoupled decay).

#include <torch/extension.h>
#include <ATen/Dispatch.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

template <typename GT, typename PL, bool NESTEROV, bool MOMENTUM>
__global__ void fused_sgd_kernel(float* __restrict__ p,
                                 PL* __restrict__ p_lowp,
                                 const GT* __restrict__ g,
                                 float* __restrict__ v,
                                 float lr, float momentum, float wd,
                                 float dampening, float gscale, long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  const long n8 = n / 8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    const long base = i * 8;
    float rp[8], rg[8], rv[8];
    V8<float>::load(p + base, rp);
    V8<GT>::load(g + base, rg);
    if (MOMENTUM) V8<float>::load(v + base, rv);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float grad = rg[k] * gscale + wd * rp[k];
      if (MOMENTUM) {
        rv[k] = momentum * rv[k] + (1.f - dampening) * grad;
        grad = NESTEROV ? grad + momentum * rv[k] : rv[k];
      }
      rp[k] -= lr * grad;
    }
    V8<float>::store(p + base, rp);
    if (MOMENTUM) V8<float>::store(v + base, rv);
    if (p_lowp != nullptr) V8<PL>::store(p_lowp + base, rp);
  }
  for (long i = n8 * 8 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float grad = float(g[i]) * gscale + wd * p[i];
    if (MOMENTUM) {
      v[i] = momentum * v[i] + (1.f - dampening) * grad;
      grad = NESTEROV ? grad + momentum * v[i] : v[i];
    }
    p[i] -= lr * grad;
    if (p_lowp != nullptr) p_lowp[i] = PL(p[i]);
  }
}

template <typename GT, typename PL, bool ADAMW>
__global__ void fused_adam_kernel(float* __restrict__ p,
                                  PL* __restrict__ p_lowp,
                                  const GT* __restrict__ g,
                                  float* __restrict__ m,
                                  float* __restrict__ v,
                                  float lr, float b1, float b2, float eps,
                                  float wd, float inv_bc1, float sqrt_bc2,
                                  float gscale, long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  const long n8 = n / 8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    const long base = i * 8;
    float rp[8], rg[8], rm[8], rv[8];
    V8<float>::load(p + base, rp);
    V8<GT>::load(g + base, rg);
    V8<float>::load(m + base, rm);
    V8<float>::load(v + base, rv);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float grad = rg[k] * gscale;
      if (ADAMW) rp[k] *= (1.f - lr * wd);
      else grad += wd * rp[k];
      rm[k] = b1 * rm[k] + (1.f - b1) * grad;
      rv[k] = b2 * rv[k] + (1.f - b2) * grad * grad;
      const float denom = sqrtf(rv[k]) / sqrt_bc2 + eps;
      rp[k] -= lr * inv_bc1 * rm[k] / denom;
    }
    V8<float>::store(p + base, rp);
    V8<float>::store(m + base, rm);
    V8<float>::store(v + base, rv);
    if (p_lowp != nullptr) V8<PL>::store(p_lowp + base, rp);
  }
  for (long i = n8 * 8 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float grad = float(g[i]) * gscale;
    if (ADAMW) p[i] *= (1.f - lr * wd);
    else grad += wd * p[i];
    m[i] = b1 * m[i] + (1.f - b1) * grad;
    v[i] = b2 * v[i] + (1.f - b2) * grad * grad;
    p[i] -= lr * inv_bc1 * m[i] / (sqrtf(v[i]) / sqrt_bc2 + eps);
    if (p_lowp != nullptr) p_lowp[i] = PL(p[i]);
  }
}

__device__ __forceinline__ void load4s(const float* p, float (&r)[4]) {
  const float4 v = *reinterpret_cast<const float4*>(p);
  r[0] = v.x; r[1] = v.y; r[2] = v.z; r[3] = v.w;
}
template <typename T16>
__device__ __forceinline__ void load4s(const T16* p, float (&r)[4]) {
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

// Per-segment sum of squares over a flat buffer.  Each block owns a
// contiguous chunk; for every segment overlapping the chunk it
// block-reduces the local sum and atomically adds to out[seg].
template <typename T>
__global__ void segment_sq_norms_kernel(const T* __restrict__ flat,
                                        const long* __restrict__ offsets,
                                        const long* __restrict__ sizes,
                                        int n_segs,
                                        float* __restrict__ out,
                                        long chunk, long total) {
  __shared__ float scratch[CMN_BLOCK / CMN_WAVE];
  const long c0 = blockIdx.x * chunk;
  const long c1 = min(c0 + chunk, total);
  if (c0 >= total) return;
  int seg = find_segment(offsets, n_segs, c0);
  while (seg < n_segs && offsets[seg] < c1) {
    const long s0 = max(c0, offsets[seg]);
    const long s1 = min(c1, offsets[seg] + sizes[seg]);
    float acc = 0.f;
    // vectorized main body (s0 is 4-aligned: chunk and slot starts are
    // multiples of 256/ALIGN elements), scalar tail
    const long n4 = s0 + ((s1 - s0) / 4) * 4;
    for (long i = s0 + threadIdx.x * 4; i < n4; i += blockDim.x * 4) {
      float r[4];
      load4s(flat + i, r);
      acc += r[0] * r[0] + r[1] * r[1] + r[2] * r[2] + r[3] * r[3];
    }
    for (long i = n4 + threadIdx.x; i < s1; i += blockDim.x) {
      const float x = float(flat[i]);
      acc += x * x;
    }
    const float tot = block_reduce_sum(acc, scratch);
    if (threadIdx.x == 0 && tot != 0.f) atomicAdd(&out[seg], tot);
    ++seg;
    __syncthreads();
  }
}

// LARS: v = mom*v + trust*lr*(g*gscale + wd*p); p -= v,
// trust = eta*||p|| / (||g||*gscale + wd*||p|| + eps) per segment.
template <typename GT, typename PL>
__global__ void lars_update_kernel(float* __restrict__ p,
                                   PL* __restrict__ p_lowp,
                                   const GT* __restrict__ g,
                                   float* __restrict__ v,
                                   const long* __restrict__ offsets,
                                   const long* __restrict__ sizes,
                                   int n_segs,
                                   const float* __restrict__ p_sqn,
                                   const float* __restrict__ g_sqn,
                                   float lr, float momentum, float wd,
                                   float eta, float eps, float gscale,
                                   long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const int seg = find_segment(offsets, n_segs, i);
    const long local = i - offsets[seg];
    if (local >= sizes[seg]) continue;  // slot padding
    const float pn = sqrtf(p_sqn[seg]);
    const float gn = sqrtf(g_sqn[seg]) * gscale;
    float trust = 1.f;
    if (pn > 0.f && gn > 0.f) trust = eta * pn / (gn + wd * pn + eps);
    const float upd = float(g[i]) * gscale + wd * p[i];
    v[i] = momentum * v[i] + trust * lr * upd;
    p[i] -= v[i];
    if (p_lowp != nullptr) p_lowp[i] = PL(p[i]);
  }
}

}  // namespace

#define DISPATCH_G(TENSOR, ...)                                              \
  CMN_DISPATCH(TENSOR.scalar_type(), "grad", [&] {                           \
    using grad_t = scalar_t;                                                 \
    return __VA_ARGS__();                                                    \
  })

void fused_sgd(torch::Tensor p, torch::Tensor p_lowp, torch::Tensor g,
               torch::Tensor v, double lr, double momentum, double wd,
               double dampening, bool nesterov, double gscale) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == at::kFloat);
  const long n = p.numel();
  const bool has_mom = momentum != 0.0;
  TORCH_CHECK(!has_mom || v.numel() == n, "momentum buffer size mismatch");
  TORCH_CHECK(g.numel() >= n, "grad buffer too small");
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = cmn_grid(n / 8 + 1, CMN_BLOCK);
  DISPATCH_G(g, [&] {
    c10::BFloat16* lowp = p_lowp.numel()
        ? (c10::BFloat16*)p_lowp.data_ptr() : nullptr;
    TORCH_CHECK(!lowp || p_lowp.scalar_type() == at::kBFloat16,
                "p_lowp must be bf16");
    auto launch = [&](auto nest, auto mom) {
      hipLaunchKernelGGL(
          (fused_sgd_kernel<grad_t, c10::BFloat16, decltype(nest)::value,
                            decltype(mom)::value>),
          dim3(grid), dim3(CMN_BLOCK), 0, stream.stream(),
          p.data_ptr<float>(), lowp, (const grad_t*)g.data_ptr(),
          has_mom ? v.data_ptr<float>() : nullptr, (float)lr,
          (float)momentum, (float)wd, (float)dampening, (float)gscale, n);
    };
    if (has_mom) {
      if (nesterov) launch(std::true_type{}, std::true_type{});
      else launch(std::false_type{}, std::true_type{});
    } else {
      launch(std::false_type{}, std::false_type{});
    }
  });
}

void fused_adam(torch::Tensor p, torch::Tensor p_lowp, torch::Tensor g,
                torch::Tensor m, torch::Tensor v, double lr, double b1,
                double b2, double eps, double wd, long step, bool adamw,
                double gscale) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == at::kFloat);
  const long n = p.numel();
  TORCH_CHECK(m.numel() == n && v.numel() == n && g.numel() >= n);
  const float inv_bc1 = 1.f / (1.f - powf((float)b1, (float)step));
  const float sqrt_bc2 = sqrtf(1.f - powf((float)b2, (float)step));
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = cmn_grid(n / 8 + 1, CMN_BLOCK);
  DISPATCH_G(g, [&] {
    c10::BFloat16* lowp = p_lowp.numel()
        ? (c10::BFloat16*)p_lowp.data_ptr() : nullptr;
    auto launch = [&](auto aw) {
      hipLaunchKernelGGL(
          (fused_adam_kernel<grad_t, c10::BFloat16, decltype(aw)::value>),
          dim3(grid), dim3(CMN_BLOCK), 0, stream.stream(),
          p.data_ptr<float>(), lowp, (const grad_t*)g.data_ptr(),
          m.data_ptr<float>(), v.data_ptr<float>(), (float)lr, (float)b1,
          (float)b2, (float)eps, (float)wd, inv_bc1, sqrt_bc2,
          (float)gscale, n);
    };
    if (adamw) launch(std::true_type{});
    else launch(std::false_type{});
  });
}

void segment_sq_norms(torch::Tensor flat, torch::Tensor offsets,
                      torch::Tensor sizes, torch::Tensor out) {
  TORCH_CHECK(flat.is_cuda() && out.scalar_type() == at::kFloat);
  const int n_segs = (int)offsets.numel();
  const long total = flat.numel();
  out.zero_();
  // chunk sized so every CU gets work; 2048 blocks max
  long chunk = (total + 2047) / 2048;
  chunk = ((chunk + 255) / 256) * 256;
  if (chunk < 2048) chunk = 2048;
  const int grid = (int)((total + chunk - 1) / chunk);
  auto stream = c10::hip::getCurrentHIPStream();
  DISPATCH_G(flat, [&] {
    hipLaunchKernelGGL((segment_sq_norms_kernel<grad_t>), dim3(grid),
                       dim3(CMN_BLOCK), 0, stream.stream(),
                       (const grad_t*)flat.data_ptr(),
                       offsets.data_ptr<long>(), sizes.data_ptr<long>(),
                       n_segs, out.data_ptr<float>(), chunk, total);
  });
}

void lars_update(torch::Tensor p, torch::Tensor p_lowp, torch::Tensor g,
                 torch::Tensor v, torch::Tensor offsets, torch::Tensor sizes,
                 torch::Tensor p_sqn, torch::Tensor g_sqn, double lr,
                 double momentum, double wd, double eta, double eps,
                 double gscale) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == at::kFloat);
  const long n = p.numel();
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = cmn_grid(n, CMN_BLOCK);
  DISPATCH_G(g, [&] {
    c10::BFloat16* lowp = p_lowp.numel()
        ? (c10::BFloat16*)p_lowp.data_ptr() : nullptr;
    hipLaunchKernelGGL((lars_update_kernel<grad_t, c10::BFloat16>),
                       dim3(grid), dim3(CMN_BLOCK), 0, stream.stream(),
                       p.data_ptr<float>(), lowp,
                       (const grad_t*)g.data_ptr(), v.data_ptr<float>(),
                       offsets.data_ptr<long>(), sizes.data_ptr<long>(),
                       (int)offsets.numel(), p_sqn.data_ptr<float>(),
                       g_sqn.data_ptr<float>(), (float)lr, (float)momentum,
                       (float)wd, (float)eta, (float)eps, (float)gscale, n);
  });
}
