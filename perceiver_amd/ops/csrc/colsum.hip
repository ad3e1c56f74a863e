// Column sums over tall bf16 matrices — bias gradients (db[c] = sum_r dy[r][c])
// and the LN dw/db reductions share this machinery.
//
// torch's generic reduce runs ~18x off the bandwidth roofline on these
// tall-skinny shapes; a first atomic-publishing version here was atomic-bound
// (blocks x C fp32 atomicAdds measured slower than the data read). Final
// shape: each block streams a strided row range at 16 B/lane into register
// partials, combines its waves through LDS, writes ONE partial row per block
// (plain stores, no zero-init needed), and a small finalize kernel reduces
// the (nblocks, C) partial matrix column-wise.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

constexpr int LANES = 64;   // x: one wave covers 64 granules = 512 columns
constexpr int ROWS = 4;     // y: rows in flight per block

__global__ void colsum_partial_kernel(const unsigned short* __restrict__ x,
                                      float* __restrict__ partial,
                                      long rows, int C) {
    // statically-indexed accumulators (runtime bounds would scratch them)
    float acc[4][8];
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[j][e] = 0.f;

    const int gpr = (C + 7) / 8;
#pragma unroll 4
    for (long r = (long)blockIdx.x * ROWS + threadIdx.y; r < rows;
         r += (long)gridDim.x * ROWS) {
        const unsigned short* row = x + r * C;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            int g = threadIdx.x + j * LANES;
            int c0 = g * 8;
            if (g < gpr) {
                if (c0 + 8 <= C) {
                    short8v v = *reinterpret_cast<const short8v*>(row + c0);
#pragma unroll
                    for (int e = 0; e < 8; ++e) acc[j][e] += bf2f((unsigned short)v[e]);
                } else {
#pragma unroll
                    for (int e = 0; e < 8; ++e)
                        if (c0 + e < C) acc[j][e] += bf2f(row[c0 + e]);
                }
            }
        }
    }

    __shared__ float red[2048];
    for (int i = threadIdx.y * LANES + threadIdx.x; i < 2048; i += LANES * ROWS)
        red[i] = 0.f;
    __syncthreads();
#pragma unroll
    for (int j = 0; j < 4; ++j) {
        int c0 = (threadIdx.x + j * LANES) * 8;
#pragma unroll
        for (int e = 0; e < 8; ++e)
            if (c0 + e < C) atomicAdd(&red[c0 + e], acc[j][e]);  // LDS only
    }
    __syncthreads();
    // partial slab layout [c>>6][part][c&63]: the finalize then streams each
    // 64-column group CONTIGUOUSLY (the row-major [part][C] layout made it
    // read 4-B columns at a 4*C stride — 16-32x DRAM amplification, and the
    // whole reason more partial blocks measured SLOWER)
    for (int c = threadIdx.y * LANES + threadIdx.x; c < C; c += LANES * ROWS)
        partial[((long)(c >> 6) * gridDim.x + blockIdx.x) * LANES + (c & 63)] = red[c];
}

// out[cb*64+lane] = sum_p partial[cb][p][lane]: one block per 64-column
// group; each (p) row of the group is a contiguous 256-B burst, y-waves
// split the parts so the streams stay deep
constexpr int FIN_ROWS = 16;

__global__ void colsum_finalize_kernel(const float* __restrict__ partial,
                                       float* __restrict__ out,
                                       int nparts, int C) {
    const float* grp = partial + (long)blockIdx.x * nparts * LANES;
    float acc = 0.f;
#pragma unroll 8
    for (int p = threadIdx.y; p < nparts; p += FIN_ROWS)
        acc += grp[(long)p * LANES + threadIdx.x];
    __shared__ float red[FIN_ROWS][LANES];
    red[threadIdx.y][threadIdx.x] = acc;
    __syncthreads();
    int c = blockIdx.x * LANES + threadIdx.x;
    if (threadIdx.y == 0 && c < C) {
        float total = red[0][threadIdx.x];
#pragma unroll
        for (int y = 1; y < FIN_ROWS; ++y) total += red[y][threadIdx.x];
        out[c] = total;
    }
}

}  // namespace

void colsum_reduce_partials(const torch::Tensor& partial, torch::Tensor& out,
                            int nparts, int C) {
    dim3 block(LANES, FIN_ROWS);
    hipLaunchKernelGGL(colsum_finalize_kernel, dim3((C + LANES - 1) / LANES), block, 0,
                       at::cuda::getCurrentCUDAStream(),
                       partial.data_ptr<float>(), out.data_ptr<float>(), nparts, C);
    HIP_CHECK_LAST();
}

torch::Tensor colsum_bf16(torch::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 && x.dim() == 2);
    x = x.contiguous();
    long rows = x.size(0);
    int C = x.size(1);
    TORCH_CHECK(C <= 2048, "colsum_bf16: C must be <= 2048");
    if (rows == 0) return torch::zeros({(long)C}, x.options().dtype(torch::kFloat32));
    static const long kRedBlocks = [] {
        const char* e = getenv("PERCEIVER_RED_BLOCKS");
        return e ? atol(e) : 1024L;
    }();
    long nblocks = std::min((rows + ROWS - 1) / ROWS, kRedBlocks);
    long cgroups = (C + LANES - 1) / LANES;
    auto partial = torch::empty({cgroups * nblocks, (long)LANES},
                                x.options().dtype(torch::kFloat32));
    auto out = torch::empty({(long)C}, x.options().dtype(torch::kFloat32));
    dim3 block(LANES, ROWS);
    hipLaunchKernelGGL(colsum_partial_kernel, dim3(nblocks), block, 0,
                       at::cuda::getCurrentCUDAStream(),
                       reinterpret_cast<const unsigned short*>(x.data_ptr()),
                       partial.data_ptr<float>(), rows, C);
    HIP_CHECK_LAST();
    colsum_reduce_partials(partial, out, (int)nblocks, C);
    return out;
}
