// Fast bf16 2-D transpose: out (K, N) = in (N, K)^T.
//
// The custom-GEMM dgrad path feeds nn.Linear weights to the B^T kernel
// transposed; torch's generic strided copy for that runs far off the
// bandwidth roofline on 2-byte elements (gather-side 2-B accesses), and it
// sits on the critical path ~80 times per training step. Classic LDS-tiled
// transpose: 64x64 tiles staged through padded LDS, 16-B coalesced loads on
// the input side, 16-B coalesced stores on the output side.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

constexpr int TDIM = 64;   // tile is TDIM x TDIM elements
constexpr int TPAD = 8;    // element pad per row breaks bank conflicts

__launch_bounds__(256)
__global__ void transpose_bf16_kernel(const unsigned short* __restrict__ in,
                                      unsigned short* __restrict__ out,
                                      int N, int K) {
    __shared__ unsigned short tile[TDIM][TDIM + TPAD];
    const int tx = threadIdx.x;             // 0..255
    const int n0 = blockIdx.x * TDIM;
    const int k0 = blockIdx.y * TDIM;

    // load: 64 rows x 64 cols = 512 granules of 16 B; 256 threads x 2
#pragma unroll
    for (int i = 0; i < 2; ++i) {
        int g = tx + i * 256;
        int row = g / 8, c8 = (g % 8) * 8;
        int n = n0 + row, k = k0 + c8;
        if (n < N) {
            if (k + 8 <= K) {
                short8v v = *reinterpret_cast<const short8v*>(in + (long)n * K + k);
#pragma unroll
                for (int e = 0; e < 8; ++e) tile[row][c8 + e] = (unsigned short)v[e];
            } else {
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    if (k + e < K) tile[row][c8 + e] = in[(long)n * K + k + e];
            }
        }
    }
    __syncthreads();
    // store: out row = k, cols = n; gather the transposed 16-B chunk from LDS
#pragma unroll
    for (int i = 0; i < 2; ++i) {
        int g = tx + i * 256;
        int row = g / 8, c8 = (g % 8) * 8;
        int k = k0 + row, n = n0 + c8;
        if (k < K) {
            if (n + 8 <= N) {
                short8v v;
#pragma unroll
                for (int e = 0; e < 8; ++e) v[e] = (short)tile[c8 + e][row];
                *reinterpret_cast<short8v*>(out + (long)k * N + n) = v;
            } else {
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    if (n + e < N) out[(long)k * N + n + e] = tile[c8 + e][row];
            }
        }
    }
}

}  // namespace

torch::Tensor transpose_bf16(torch::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 && x.dim() == 2,
                "transpose_bf16: 2-D cuda bf16 only");
    x = x.contiguous();
    long N = x.size(0), K = x.size(1);
    auto out = torch::empty({K, N}, x.options());
    dim3 grid((N + TDIM - 1) / TDIM, (K + TDIM - 1) / TDIM);
    hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(),
                       reinterpret_cast<const unsigned short*>(x.data_ptr()),
                       reinterpret_cast<unsigned short*>(out.data_ptr()),
                       (int)N, (int)K);
    HIP_CHECK_LAST();
    return out;
}
