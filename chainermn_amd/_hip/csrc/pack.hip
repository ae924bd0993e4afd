// Gradient pack/unpack + flat cast kernels (gfx950).
//
// Rebuilds of the reference's CuPy JIT kernels:
//  K1 batched pack   (_memory_utility.py ~L200-280): gather param grads
//     into a flat wire buffer, fused dtype cast, zero-filled slot padding
//  K2 batched unpack (_memory_utility.py ~L280-350): scatter flat -> grads
//     with fused ×scale and cast-back — one HBM round-trip
//  K3 flat cast      (pure_nccl_communicator.py ~L230-300): arena-path
//     cast with optional fused scale (subsumes the reference's separate
//     cast and ×1/size kernels)
//
// All are HBM-bandwidth-bound gathers/scatters: 8 elems/lane, uint4/float4
// transactions, grid-stride with capped grid.  Param slots in the flat
// buffer are 64-element aligned (ops.ALIGN) so an 8-wide vector never
// crosses a parameter boundary.

#include <torch/extension.h>
#include <ATen/Dispatch.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

template <typename SRC, typename DST>
__global__ void batched_pack_kernel(const long* __restrict__ ptrs,
                                    const long* __restrict__ sizes,
                                    const long* __restrict__ offsets,
                                    int n_params,
                                    DST* __restrict__ out,
                                    long total8) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = blockIdx.x * (long)blockDim.x + threadIdx.x; v < total8;
       v += stride) {
    const long base = v * 8;
    const int seg = find_segment(offsets, n_params, base);
    const long local = base - offsets[seg];
    const long size = sizes[seg];
    const SRC* __restrict__ src = reinterpret_cast<const SRC*>(ptrs[seg]);
    float r[8];
    if (local + 8 <= size) {
      V8<SRC>::load(src + local, r);
    } else {
#pragma unroll
      for (int k = 0; k < 8; ++k)
        r[k] = (local + k < size) ? float(src[local + k]) : 0.f;
    }
    V8<DST>::store(out + base, r);
  }
}

template <typename SRC, typename DST>
__global__ void batched_unpack_kernel(const long* __restrict__ ptrs,
                                      const long* __restrict__ sizes,
                                      const long* __restrict__ offsets,
                                      int n_params,
                                      const SRC* __restrict__ in,
                                      float scale,
                                      long total8) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = blockIdx.x * (long)blockDim.x + threadIdx.x; v < total8;
       v += stride) {
    const long base = v * 8;
    const int seg = find_segment(offsets, n_params, base);
    const long local = base - offsets[seg];
    const long size = sizes[seg];
    if (local >= size) continue;  // pad region
    DST* __restrict__ dst = reinterpret_cast<DST*>(ptrs[seg]);
    float r[8];
    V8<SRC>::load(in + base, r);
#pragma unroll
    for (int k = 0; k < 8; ++k) r[k] *= scale;
    if (local + 8 <= size) {
      V8<DST>::store(dst + local, r);
    } else {
#pragma unroll
      for (int k = 0; k < 8; ++k)
        if (local + k < size) dst[local + k] = DST(r[k]);
    }
  }
}

template <typename SRC, typename DST>
__global__ void cast_flat_kernel(const SRC* __restrict__ src,
                                 DST* __restrict__ dst, float scale, long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  const long n8 = n / 8;
  for (long v = blockIdx.x * (long)blockDim.x + threadIdx.x; v < n8;
       v += stride) {
    float r[8];
    V8<SRC>::load(src + v * 8, r);
#pragma unroll
    for (int k = 0; k < 8; ++k) r[k] *= scale;
    V8<DST>::store(dst + v * 8, r);
  }
  // scalar tail
  for (long i = n8 * 8 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride)
    dst[i] = DST(float(src[i]) * scale);
}

}  // namespace

#define DISPATCH_PAIR(SRC_T, DST_T, ...)                                     \
  CMN_DISPATCH(SRC_T, "src", [&] {                                           \
    using src_t = scalar_t;                                                  \
    return CMN_DISPATCH(DST_T, "dst", [&] {                                  \
      using dst_t = scalar_t;                                                \
      return __VA_ARGS__();                                                  \
    });                                                                      \
  })

void batched_pack(torch::Tensor ptrs, torch::Tensor sizes,
                  torch::Tensor offsets, torch::Tensor out,
                  torch::ScalarType src_dtype) {
  TORCH_CHECK(out.is_cuda() && ptrs.is_cuda(), "batched_pack: GPU only");
  const int n_params = (int)ptrs.numel();
  const long total8 = out.numel() / 8;
  auto stream = c10::hip::getCurrentHIPStream();
  DISPATCH_PAIR(src_dtype, out.scalar_type(), [&] {
    hipLaunchKernelGGL((batched_pack_kernel<src_t, dst_t>),
                       dim3(cmn_grid(total8, CMN_BLOCK)), dim3(CMN_BLOCK), 0,
                       stream.stream(), ptrs.data_ptr<long>(),
                       sizes.data_ptr<long>(), offsets.data_ptr<long>(),
                       n_params, (dst_t*)out.data_ptr(), total8);
  });
}

void batched_unpack(torch::Tensor ptrs, torch::Tensor sizes,
                    torch::Tensor offsets, torch::Tensor in, double scale,
                    torch::ScalarType dst_dtype) {
  TORCH_CHECK(in.is_cuda() && ptrs.is_cuda(), "batched_unpack: GPU only");
  const int n_params = (int)ptrs.numel();
  const long total8 = in.numel() / 8;
  auto stream = c10::hip::getCurrentHIPStream();
  DISPATCH_PAIR(in.scalar_type(), dst_dtype, [&] {
    hipLaunchKernelGGL((batched_unpack_kernel<src_t, dst_t>),
                       dim3(cmn_grid(total8, CMN_BLOCK)), dim3(CMN_BLOCK), 0,
                       stream.stream(), ptrs.data_ptr<long>(),
                       sizes.data_ptr<long>(), offsets.data_ptr<long>(),
                       n_params, (const src_t*)in.data_ptr(), (float)scale,
                       total8);
  });
}

void cast_flat(torch::Tensor src, torch::Tensor dst, double scale) {
  TORCH_CHECK(src.is_cuda() && dst.is_cuda(), "cast_flat: GPU only");
  TORCH_CHECK(src.numel() == dst.numel(), "cast_flat: size mismatch");
  TORCH_CHECK(src.is_contiguous() && dst.is_contiguous());
  const long n = src.numel();
  auto stream = c10::hip::getCurrentHIPStream();
  DISPATCH_PAIR(src.scalar_type(), dst.scalar_type(), [&] {
    hipLaunchKernelGGL((cast_flat_kernel<src_t, dst_t>),
                       dim3(cmn_grid(n / 8 + 1, CMN_BLOCK)), dim3(CMN_BLOCK),
                       0, stream.stream(), (const src_t*)src.data_ptr(),
                       (dst_t*)dst.data_ptr(), (float)scale, n);
  });
}
