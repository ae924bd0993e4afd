// Channels-last 3x3/stride-2/pad-1 max-pooling (gfx950) — the ResNet stem
// pool.  torch's max_pool_backward_nhwc runs ~6x over the bandwidth bound
// on MI355X (1.24 ms at batch 512, see profiles/).
//
// Forward stores a per-element window code (0..8, uint8) so backward is a
// GATHER over the <=4 windows containing each input cell — no atomics, no
// zero-init, fully coalesced 8-channel uint4 lanes.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

template <typename T>
__global__ void maxpool3x3s2_fwd_cl_kernel(const T* __restrict__ x,
                                           T* __restrict__ y,
                                           unsigned char* __restrict__ idx,
                                           int C, int IH, int IW, int OH,
                                           int OW, long total_vec) {
  const int cv = C / 8;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = blockIdx.x * (long)blockDim.x + threadIdx.x; v < total_vec;
       v += stride) {
    const int c0 = (int)(v % cv) * 8;
    long rest = v / cv;
    const int ow = (int)(rest % OW);
    rest /= OW;
    const int oh = (int)(rest % OH);
    const long n = rest / OH;
    const int ih0 = oh * 2 - 1;
    const int iw0 = ow * 2 - 1;
    float best[8];
    int bidx[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      best[k] = -3.4e38f;
      bidx[k] = 0;
    }
    for (int kh = 0; kh < 3; ++kh) {
      const int ih = ih0 + kh;
      if (ih < 0 || ih >= IH) continue;
      for (int kw = 0; kw < 3; ++kw) {
        const int iw = iw0 + kw;
        if (iw < 0 || iw >= IW) continue;
        const long base = (((n * IH + ih) * IW) + iw) * C + c0;
        float r[8];
        V8<T>::load(x + base, r);
        const int code = kh * 3 + kw;
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          if (r[k] > best[k]) {  // strict > keeps the FIRST max (torch)
            best[k] = r[k];
            bidx[k] = code;
          }
        }
      }
    }
    const long obase = v * 8;
    V8<T>::store(y + obase, best);
    uint2 packed;
    packed.x = (unsigned)bidx[0] | ((unsigned)bidx[1] << 8)
               | ((unsigned)bidx[2] << 16) | ((unsigned)bidx[3] << 24);
    packed.y = (unsigned)bidx[4] | ((unsigned)bidx[5] << 8)
               | ((unsigned)bidx[6] << 16) | ((unsigned)bidx[7] << 24);
    *reinterpret_cast<uint2*>(idx + obase) = packed;
  }
}

template <typename T>
__global__ void maxpool3x3s2_bwd_cl_kernel(const T* __restrict__ gy,
                                           const unsigned char* __restrict__
                                               idx,
                                           T* __restrict__ gx,
                                           int C, int IH, int IW, int OH,
                                           int OW, long total_vec) {
  const int cv = C / 8;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = blockIdx.x * (long)blockDim.x + threadIdx.x; v < total_vec;
       v += stride) {
    const int c0 = (int)(v % cv) * 8;
    long rest = v / cv;
    const int iw = (int)(rest % IW);
    rest /= IW;
    const int ih = (int)(rest % IH);
    const long n = rest / IH;
    float acc[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) acc[k] = 0.f;
    for (int kh = 0; kh < 3; ++kh) {
      const int t = ih + 1 - kh;
      if (t < 0 || (t & 1)) continue;
      const int oh = t >> 1;
      if (oh >= OH) continue;
      for (int kw = 0; kw < 3; ++kw) {
        const int s = iw + 1 - kw;
        if (s < 0 || (s & 1)) continue;
        const int ow = s >> 1;
        if (ow >= OW) continue;
        const long obase = (((n * OH + oh) * OW) + ow) * C + c0;
        float g[8];
        V8<T>::load(gy + obase, g);
        const uint2 packed = *reinterpret_cast<const uint2*>(idx + obase);
        const unsigned code = (unsigned)(kh * 3 + kw);
        const unsigned pk[2] = {packed.x, packed.y};
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          const unsigned b = (pk[k >> 2] >> ((k & 3) * 8)) & 0xffu;
          if (b == code) acc[k] += g[k];
        }
      }
    }
    V8<T>::store(gx + v * 8, acc);
  }
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor> maxpool3x3s2_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4
              && x.is_contiguous(at::MemoryFormat::ChannelsLast)
              && x.size(1) % 8 == 0,
              "maxpool3x3s2_fwd: channels_last 4-D, C % 8 == 0");
  const long N = x.size(0);
  const int C = (int)x.size(1), IH = (int)x.size(2), IW = (int)x.size(3);
  const int OH = (IH + 2 - 3) / 2 + 1;
  const int OW = (IW + 2 - 3) / 2 + 1;
  auto y = torch::empty({N, C, OH, OW},
                        x.options().memory_format(
                            at::MemoryFormat::ChannelsLast));
  auto idx = torch::empty({N * OH * OW * C},
                          x.options().dtype(at::kByte));
  const long total_vec = N * (long)OH * OW * C / 8;
  auto stream = c10::hip::getCurrentHIPStream();
  CMN_DISPATCH(x.scalar_type(), "maxpool_fwd", [&] {
    hipLaunchKernelGGL((maxpool3x3s2_fwd_cl_kernel<scalar_t>),
                       dim3(cmn_grid(total_vec, CMN_BLOCK)),
                       dim3(CMN_BLOCK), 0, stream.stream(),
                       (const scalar_t*)x.data_ptr(),
                       (scalar_t*)y.data_ptr(),
                       idx.data_ptr<unsigned char>(), C, IH, IW, OH, OW,
                       total_vec);
  });
  return {y, idx};
}

torch::Tensor maxpool3x3s2_bwd(torch::Tensor gy, torch::Tensor idx,
                               long IH, long IW) {
  TORCH_CHECK(gy.is_cuda() && gy.dim() == 4
              && gy.is_contiguous(at::MemoryFormat::ChannelsLast));
  const long N = gy.size(0);
  const int C = (int)gy.size(1), OH = (int)gy.size(2), OW = (int)gy.size(3);
  auto gx = torch::empty({N, C, IH, IW},
                         gy.options().memory_format(
                             at::MemoryFormat::ChannelsLast));
  const long total_vec = N * IH * IW * C / 8;
  auto stream = c10::hip::getCurrentHIPStream();
  CMN_DISPATCH(gy.scalar_type(), "maxpool_bwd", [&] {
    hipLaunchKernelGGL((maxpool3x3s2_bwd_cl_kernel<scalar_t>),
                       dim3(cmn_grid(total_vec, CMN_BLOCK)),
                       dim3(CMN_BLOCK), 0, stream.stream(),
                       (const scalar_t*)gy.data_ptr(),
                       idx.data_ptr<unsigned char>(),
                       (scalar_t*)gx.data_ptr(), C, (int)IH, (int)IW, OH,
                       OW, total_vec);
  });
  return gx;
}
