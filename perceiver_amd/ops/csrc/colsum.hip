// Column sums over tall bf16 matrices — bias gradients (db[c] = sum_r dy[r][c])
// and the LN dw/db finalize share this machinery.
//
// torch's generic reduce runs ~18x off the bandwidth roofline on these
// tall-skinny shapes. Design history: an atomic-publishing version was
// atomic-bound; a block-combined (LDS atomics + one partial row per block)
// version measured ~1.3 TB/s — the per-block epilogue (zero/combine/write)
// throttled how many blocks could be worth launching. Final shape: each WAVE
// owns whole rows (grid-stride), accumulates per-lane fp32 partials in
// registers, and writes its OWN partial row — no LDS, no atomics, no
// barrier. The slab is tiled [c>>6][part][c&63] so the finalize kernel
// streams 256-B bursts (the row-major slab cost it 16-32x DRAM
// amplification reading 4-B columns at a 4*C stride).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

constexpr int LANES = 64;
constexpr int WPB = 4;       // waves per block
constexpr int CS_MAXC = 2048;

// CHUNKS = ceil(C / 512); lane covers 8 contiguous elems per chunk.
template <int CHUNKS>
__launch_bounds__(LANES* WPB, CHUNKS >= 3 ? 2 : 4)
__global__ void colsum_partial_kernel(const unsigned short* __restrict__ x,
                                      float* __restrict__ partial,
                                      long rows, int C) {
    const int lane = threadIdx.x % 64;
    const long wave0 = (long)blockIdx.x * WPB + threadIdx.x / 64;
    const long wstride = (long)gridDim.x * WPB;
    const long nparts = (long)gridDim.x * WPB;

    float acc[CHUNKS][8];
#pragma unroll
    for (int i = 0; i < CHUNKS; ++i)
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[i][e] = 0.f;

    for (long r = wave0; r < rows; r += wstride) {
        const unsigned short* row = x + r * C;
#pragma unroll
        for (int i = 0; i < CHUNKS; ++i) {
            int c0 = lane * 8 + i * 512;
            if (c0 + 8 <= C) {
                short8v v = *reinterpret_cast<const short8v*>(row + c0);
#pragma unroll
                for (int e = 0; e < 8; ++e) acc[i][e] += bf2f((unsigned short)v[e]);
            } else if (c0 < C) {
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    if (c0 + e < C) acc[i][e] += bf2f(row[c0 + e]);
            }
        }
    }

    // wave-private tiled partial row; whole-granule stores per chunk
#pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
        int c0 = lane * 8 + i * 512;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            int c = c0 + e;
            if (c < C)
                partial[((long)(c >> 6) * nparts + wave0) * LANES + (c & 63)] = acc[i][e];
        }
    }
}

// out[cb*64+lane] = sum_p partial[cb][p][lane]: one block per 64-column
// group; each (p) row of the group is a contiguous 256-B burst, y-waves
// split the parts so the streams stay deep
constexpr int FIN_ROWS = 16;

// OUT_BF16: emit bf16 directly — the fp32->bf16 `.to()` after every finalize
// was ~200 tiny launch-bound conversion kernels per MLM step (~1.6 ms)
template <bool OUT_BF16>
__global__ void colsum_finalize_kernel(const float* __restrict__ partial,
                                       void* __restrict__ out,
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
        if (OUT_BF16)
            reinterpret_cast<unsigned short*>(out)[c] = f2bf(total);
        else
            reinterpret_cast<float*>(out)[c] = total;
    }
}

}  // namespace

void colsum_reduce_partials(const torch::Tensor& partial, torch::Tensor& out,
                            int nparts, int C) {
    dim3 block(LANES, FIN_ROWS);
    if (out.scalar_type() == torch::kBFloat16)
        hipLaunchKernelGGL((colsum_finalize_kernel<true>), dim3((C + LANES - 1) / LANES),
                           block, 0, at::cuda::getCurrentCUDAStream(),
                           partial.data_ptr<float>(), out.data_ptr(), nparts, C);
    else
        hipLaunchKernelGGL((colsum_finalize_kernel<false>), dim3((C + LANES - 1) / LANES),
                           block, 0, at::cuda::getCurrentCUDAStream(),
                           partial.data_ptr<float>(), out.data_ptr(), nparts, C);
    HIP_CHECK_LAST();
}

torch::Tensor colsum_bf16(torch::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 && x.dim() == 2);
    x = x.contiguous();
    long rows = x.size(0);
    int C = x.size(1);
    TORCH_CHECK(C <= CS_MAXC, "colsum_bf16: C must be <= 2048");
    if (rows == 0) return torch::zeros({(long)C}, x.options());
    static const long kRedBlocks = [] {
        const char* e = getenv("PERCEIVER_RED_BLOCKS");
        return e ? atol(e) : 1024L;
    }();
    // row-adaptive block count (measured: 512 best for <=32k rows, 1024 for
    // 64k+; more blocks shrink rows/wave until latency dominates)
    long nblocks = std::min(std::max(rows / 32, (long)256), kRedBlocks);
    nblocks = std::min(nblocks, (rows + WPB - 1) / WPB);
    long nparts = nblocks * WPB;  // one partial row per WAVE
    long cgroups = (C + LANES - 1) / LANES;
    auto partial = torch::empty({cgroups * nparts, (long)LANES},
                                x.options().dtype(torch::kFloat32));
    auto out = torch::empty({(long)C}, x.options());  // bf16, rounded in-kernel
    const unsigned short* xp = reinterpret_cast<const unsigned short*>(x.data_ptr());
    auto stream = at::cuda::getCurrentCUDAStream();
    int chunks = (C + 511) / 512;
    if (chunks == 1)
        hipLaunchKernelGGL((colsum_partial_kernel<1>), dim3(nblocks), dim3(LANES * WPB), 0,
                           stream, xp, partial.data_ptr<float>(), rows, C);
    else if (chunks == 2)
        hipLaunchKernelGGL((colsum_partial_kernel<2>), dim3(nblocks), dim3(LANES * WPB), 0,
                           stream, xp, partial.data_ptr<float>(), rows, C);
    else if (chunks == 3)
        hipLaunchKernelGGL((colsum_partial_kernel<3>), dim3(nblocks), dim3(LANES * WPB), 0,
                           stream, xp, partial.data_ptr<float>(), rows, C);
    else
        hipLaunchKernelGGL((colsum_partial_kernel<4>), dim3(nblocks), dim3(LANES * WPB), 0,
                           stream, xp, partial.data_ptr<float>(), rows, C);
    HIP_CHECK_LAST();
    colsum_reduce_partials(partial, out, (int)nparts, C);
    return out;
}
