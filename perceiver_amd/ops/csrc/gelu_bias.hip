// Fused bias + exact-erf GELU, forward and backward (SURVEY.md §2.3 K6 epilogue
// piece). Memory-bound: bf16 I/O vectorized 8 elements/lane (guideline 13),
// grid-stride loop capped at ~2048 blocks (guideline 11).
//
//   fwd:  y = gelu(x + b)          x: (rows, C) bf16, b: (C) bf16 or absent
//   bwd:  dx = dy * gelu'(x + b)   (db falls out of a matmul-free column sum done
//                                   by the caller on dx_pre when bias is present)
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

__global__ void gelu_bias_fwd_kernel_bf16(
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ bias,  // may be null
    unsigned short* __restrict__ y,
    long total, int C) {
    long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    long stride = (long)gridDim.x * blockDim.x * 8;
    for (long i = i0; i + 8 <= total; i += stride) {
        short8v xv = *reinterpret_cast<const short8v*>(x + i);
        // one modulo per granule: i and C are both multiples of 8, so the
        // 8-element span never wraps the bias row
        const int c0 = bias ? (int)(i % C) : 0;
        short8v yv;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float v = bf2f((unsigned short)xv[j]);
            if (bias) v += bf2f(bias[c0 + j]);
            yv[j] = (short)f2bf(gelu_f(v));
        }
        *reinterpret_cast<short8v*>(y + i) = yv;
    }
}

__global__ void gelu_bias_bwd_kernel_bf16(
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ bias,
    const unsigned short* __restrict__ dy,
    unsigned short* __restrict__ dx,
    long total, int C) {
    long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    long stride = (long)gridDim.x * blockDim.x * 8;
    for (long i = i0; i + 8 <= total; i += stride) {
        short8v xv = *reinterpret_cast<const short8v*>(x + i);
        short8v gv = *reinterpret_cast<const short8v*>(dy + i);
        const int c0 = bias ? (int)(i % C) : 0;
        short8v ov;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float v = bf2f((unsigned short)xv[j]);
            if (bias) v += bf2f(bias[c0 + j]);
            float g = bf2f((unsigned short)gv[j]);
            ov[j] = (short)f2bf(g * gelu_grad_f(v));
        }
        *reinterpret_cast<short8v*>(dx + i) = ov;
    }
}

}  // namespace

torch::Tensor gelu_bias_fwd(torch::Tensor x, c10::optional<torch::Tensor> bias) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous());
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "gelu_bias_fwd: bf16 only");
    auto y = torch::empty_like(x);
    long total = x.numel();
    int C = x.size(-1);
    TORCH_CHECK(C % 8 == 0, "inner dim must be a multiple of 8");
    const unsigned short* bptr = nullptr;
    if (bias.has_value() && bias->defined()) {
        TORCH_CHECK(bias->is_contiguous() && bias->numel() == C);
        bptr = reinterpret_cast<const unsigned short*>(bias->data_ptr());
    }
    int threads = 256;
    long blocks = std::max<long>(1, std::min<long>((total / 8 + threads - 1) / threads, 2048));
    hipLaunchKernelGGL(gelu_bias_fwd_kernel_bf16, dim3(blocks), dim3(threads), 0,
                       at::cuda::getCurrentCUDAStream(),
                       reinterpret_cast<const unsigned short*>(x.data_ptr()), bptr,
                       reinterpret_cast<unsigned short*>(y.data_ptr()), total, C);
    HIP_CHECK_LAST();
    return y;
}

torch::Tensor gelu_bias_bwd(torch::Tensor x, c10::optional<torch::Tensor> bias, torch::Tensor dy) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "gelu_bias_bwd: bf16 only");
    auto dx = torch::empty_like(x);
    long total = x.numel();
    int C = x.size(-1);
    TORCH_CHECK(C % 8 == 0, "inner dim must be a multiple of 8");
    const unsigned short* bptr = nullptr;
    if (bias.has_value() && bias->defined()) {
        bptr = reinterpret_cast<const unsigned short*>(bias->data_ptr());
    }
    int threads = 256;
    long blocks = std::max<long>(1, std::min<long>((total / 8 + threads - 1) / threads, 2048));
    hipLaunchKernelGGL(gelu_bias_bwd_kernel_bf16, dim3(blocks), dim3(threads), 0,
                       at::cuda::getCurrentCUDAStream(),
                       reinterpret_cast<const unsigned short*>(x.data_ptr()), bptr,
                       reinterpret_cast<const unsigned short*>(dy.data_ptr()),
                       reinterpret_cast<unsigned short*>(dx.data_ptr()), total, C);
    HIP_CHECK_LAST();
    return dx;
}
