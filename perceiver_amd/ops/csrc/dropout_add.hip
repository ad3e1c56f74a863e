// Fused residual dropout-add for CDNA4 (gfx950) — part of the MLM elementwise
// tail (residual adds and dropout masks were
// ~12 ms/step of separate memory-bound kernels).
//
//   out = residual + dropout(x, p)        (train-mode scaling by 1/(1-p))
//
// One kernel instead of torch's {philox mask, mul, add} chain: one read of x,
// one of residual, one write. The mask is never materialized — the backward
// regenerates it from the same counter-hash RNG (seed, element index), so
//   dx = keep ? dy/(1-p) : 0,   dresidual = dy (the incoming tensor, no copy).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) unsigned short ushort8;

__global__ void dropout_add_fwd_kernel(const unsigned short* __restrict__ x,
                                       const unsigned short* __restrict__ res,
                                       unsigned short* __restrict__ out,
                                       long n8, float p, float inv_keep,
                                       unsigned long long seed) {
    const unsigned int thresh = (unsigned int)(p * 4294967296.0);
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < n8; i += stride) {
        ushort8 xv = *reinterpret_cast<const ushort8*>(x + i * 8);
        ushort8 rv = *reinterpret_cast<const ushort8*>(res + i * 8);
        ushort8 ov;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            bool keep = rng_hash(seed, 0, (int)(i & 0x7fffffff), (int)((i >> 31) * 8 + e)) >= thresh;
            float v = keep ? bf2f(xv[e]) * inv_keep : 0.f;
            ov[e] = f2bf(v + bf2f(rv[e]));
        }
        *reinterpret_cast<ushort8*>(out + i * 8) = ov;
    }
}

__global__ void dropout_add_bwd_kernel(const unsigned short* __restrict__ dy,
                                       unsigned short* __restrict__ dx,
                                       long n8, float p, float inv_keep,
                                       unsigned long long seed) {
    const unsigned int thresh = (unsigned int)(p * 4294967296.0);
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < n8; i += stride) {
        ushort8 gv = *reinterpret_cast<const ushort8*>(dy + i * 8);
        ushort8 ov;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            bool keep = rng_hash(seed, 0, (int)(i & 0x7fffffff), (int)((i >> 31) * 8 + e)) >= thresh;
            ov[e] = keep ? f2bf(bf2f(gv[e]) * inv_keep) : (unsigned short)0;
        }
        *reinterpret_cast<ushort8*>(dx + i * 8) = ov;
    }
}

}  // namespace

torch::Tensor dropout_add_fwd(torch::Tensor x, torch::Tensor res, double p, int64_t seed) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16, "dropout_add: bf16 CUDA only");
    TORCH_CHECK(x.is_contiguous() && res.is_contiguous() && x.numel() == res.numel());
    TORCH_CHECK(x.numel() % 8 == 0, "dropout_add: numel must be a multiple of 8");
    auto out = torch::empty_like(x);
    long n8 = x.numel() / 8;
    long blocks = std::min((n8 + 255) / 256, (long)2048);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(dropout_add_fwd_kernel, dim3(blocks), dim3(256), 0, stream,
                       reinterpret_cast<const unsigned short*>(x.data_ptr()),
                       reinterpret_cast<const unsigned short*>(res.data_ptr()),
                       reinterpret_cast<unsigned short*>(out.data_ptr()),
                       n8, (float)p, 1.0f / (1.0f - (float)p), (unsigned long long)seed);
    HIP_CHECK_LAST();
    return out;
}

torch::Tensor dropout_add_bwd(torch::Tensor dy, double p, int64_t seed) {
    TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16);
    dy = dy.contiguous();
    auto dx = torch::empty_like(dy);
    long n8 = dy.numel() / 8;
    long blocks = std::min((n8 + 255) / 256, (long)2048);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(dropout_add_bwd_kernel, dim3(blocks), dim3(256), 0, stream,
                       reinterpret_cast<const unsigned short*>(dy.data_ptr()),
                       reinterpret_cast<unsigned short*>(dx.data_ptr()),
                       n8, (float)p, 1.0f / (1.0f - (float)p), (unsigned long long)seed);
    HIP_CHECK_LAST();
    return dx;
}
