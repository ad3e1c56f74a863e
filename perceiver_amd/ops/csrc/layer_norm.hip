// Fused bf16 LayerNorm forward/backward for CDNA4 (the LN half of SURVEY.md §2.3 K6).
// Memory-bound: one wave64 per row, ushort8 vector loads, fp32 accumulation,
// cross-lane shfl_xor reductions; saves per-row mean/rstd for the backward.
// Register caching uses compile-time CHUNKS (static indexing — guide §5.4 rule 20);
// the backward computes dx AND the dW/db per-block partials in one fused pass
// (tiled partial slab + the colsum finalize kernel).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

void colsum_reduce_partials(const torch::Tensor& partial, torch::Tensor& out,
                            int nparts, int C);

namespace {

constexpr int DW_LANES = 64;  // partial-slab tile width (shared with colsum)

DEVINL float wave_sum(float x) {
#pragma unroll
    for (int m = 1; m < 64; m <<= 1) x += __shfl_xor(x, m, 64);
    return x;
}

// CHUNKS = ceil(C / 512); lane handles 8 contiguous elems per chunk.
template <int CHUNKS>
__global__ void ln_fwd_kernel(const unsigned short* __restrict__ x,
                              const unsigned short* __restrict__ w,
                              const unsigned short* __restrict__ b,
                              unsigned short* __restrict__ y,
                              float* __restrict__ mean_out, float* __restrict__ rstd_out,
                              long rows, int C, float eps) {
    long row = (long)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    int lane = threadIdx.x % 64;
    if (row >= rows) return;
    const unsigned short* xrow = x + row * C;
    unsigned short* yrow = y + row * C;

    float vals[CHUNKS][8];
    float sum = 0.f, sumsq = 0.f;
#pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
        int c0 = lane * 8 + i * 512;
#pragma unroll
        for (int e = 0; e < 8; ++e) vals[i][e] = 0.f;
        if (c0 + 8 <= C) {
            short8v v = *reinterpret_cast<const short8v*>(xrow + c0);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                float f = bf2f((unsigned short)v[e]);
                vals[i][e] = f;
                sum += f;
                sumsq += f * f;
            }
        } else if (c0 < C) {
#pragma unroll
            for (int e = 0; e < 8; ++e) {   // static index: a runtime bound
                if (c0 + e < C) {           // would push vals[] to scratch
                    float f = bf2f(xrow[c0 + e]);
                    vals[i][e] = f;
                    sum += f;
                    sumsq += f * f;
                }
            }
        }
    }
    sum = wave_sum(sum);
    sumsq = wave_sum(sumsq);
    float mean = sum / C;
    float var = sumsq / C - mean * mean;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (lane == 0) {
        mean_out[row] = mean;
        rstd_out[row] = rstd;
    }

#pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
        int c0 = lane * 8 + i * 512;
        if (c0 + 8 <= C) {
            short8v wv = *reinterpret_cast<const short8v*>(w + c0);
            short8v o;
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                float bb = b ? bf2f(b[c0 + e]) : 0.f;
                o[e] = (short)f2bf((vals[i][e] - mean) * rstd * bf2f((unsigned short)wv[e]) + bb);
            }
            *reinterpret_cast<short8v*>(yrow + c0) = o;
        } else if (c0 < C) {
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                if (c0 + e < C) {
                    float bb = b ? bf2f(b[c0 + e]) : 0.f;
                    yrow[c0 + e] = f2bf((vals[i][e] - mean) * rstd * bf2f(w[c0 + e]) + bb);
                }
            }
        }
    }
}

template <int CHUNKS>
__global__ void ln_bwd_dx_kernel(const unsigned short* __restrict__ dy,
                                 const unsigned short* __restrict__ x,
                                 const unsigned short* __restrict__ w,
                                 const float* __restrict__ mean_in,
                                 const float* __restrict__ rstd_in,
                                 unsigned short* __restrict__ dx,
                                 long rows, int C) {
    long row = (long)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    int lane = threadIdx.x % 64;
    if (row >= rows) return;
    const unsigned short* dyrow = dy + row * C;
    const unsigned short* xrow = x + row * C;
    unsigned short* dxrow = dx + row * C;
    float mean = mean_in[row], rstd = rstd_in[row];

    float g[CHUNKS][8], xh[CHUNKS][8];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
        int c0 = lane * 8 + i * 512;
#pragma unroll
        for (int e = 0; e < 8; ++e) { g[i][e] = 0.f; xh[i][e] = 0.f; }
        if (c0 + 8 <= C) {
            short8v dyv = *reinterpret_cast<const short8v*>(dyrow + c0);
            short8v xv = *reinterpret_cast<const short8v*>(xrow + c0);
            short8v wv = *reinterpret_cast<const short8v*>(w + c0);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                float gg = bf2f((unsigned short)dyv[e]) * bf2f((unsigned short)wv[e]);
                float xhat = (bf2f((unsigned short)xv[e]) - mean) * rstd;
                g[i][e] = gg;
                xh[i][e] = xhat;
                s1 += gg;
                s2 += gg * xhat;
            }
        } else if (c0 < C) {
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                if (c0 + e < C) {
                    float gg = bf2f(dyrow[c0 + e]) * bf2f(w[c0 + e]);
                    float xhat = (bf2f(xrow[c0 + e]) - mean) * rstd;
                    g[i][e] = gg;
                    xh[i][e] = xhat;
                    s1 += gg;
                    s2 += gg * xhat;
                }
            }
        }
    }
    s1 = wave_sum(s1) / C;
    s2 = wave_sum(s2) / C;

#pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
        int c0 = lane * 8 + i * 512;
        if (c0 + 8 <= C) {
            short8v o;
#pragma unroll
            for (int e = 0; e < 8; ++e)
                o[e] = (short)f2bf(rstd * (g[i][e] - s1 - xh[i][e] * s2));
            *reinterpret_cast<short8v*>(dxrow + c0) = o;
        } else if (c0 < C) {
#pragma unroll
            for (int e = 0; e < 8; ++e)
                if (c0 + e < C)
                    dxrow[c0 + e] = f2bf(rstd * (g[i][e] - s1 - xh[i][e] * s2));
        }
    }
}

// Fused backward: dx AND the dw/db per-block partials in ONE pass over
// dy/x. The separate dwdb kernel re-read the same 2*rows*C bf16 and paid a
// per-row mean/rstd load chain — ~55 us/call on the MLM shapes where the
// dx-only kernel runs the identical traffic in ~21 us. One wave per row,
// grid-stride row loop, per-lane fp32 dw/db accumulators, LDS combine,
// tiled partial slab (same [c>>6][part][c&63] layout + finalize as colsum).
template <int CHUNKS>
__launch_bounds__(256, CHUNKS >= 3 ? 2 : 4)  // wide rows need >128 VGPRs
__global__ void ln_bwd_fused_kernel(const unsigned short* __restrict__ dy,
                                    const unsigned short* __restrict__ x,
                                    const unsigned short* __restrict__ w,
                                    const float* __restrict__ mean_in,
                                    const float* __restrict__ rstd_in,
                                    unsigned short* __restrict__ dx,
                                    float* __restrict__ partial,
                                    long rows, int C) {
    const int lane = threadIdx.x % 64;
    const int wpb = blockDim.x / 64;
    const long wave0 = (long)blockIdx.x * wpb + threadIdx.x / 64;
    const long wstride = (long)gridDim.x * wpb;

    float dwacc[CHUNKS][8], dbacc[CHUNKS][8];
#pragma unroll
    for (int i = 0; i < CHUNKS; ++i)
#pragma unroll
        for (int e = 0; e < 8; ++e) { dwacc[i][e] = 0.f; dbacc[i][e] = 0.f; }

    for (long row = wave0; row < rows; row += wstride) {
        const unsigned short* dyrow = dy + row * C;
        const unsigned short* xrow = x + row * C;
        unsigned short* dxrow = dx + row * C;
        float mean = mean_in[row], rstd = rstd_in[row];

        float g[CHUNKS][8], xh[CHUNKS][8];
        float s1 = 0.f, s2 = 0.f;
#pragma unroll
        for (int i = 0; i < CHUNKS; ++i) {
            int c0 = lane * 8 + i * 512;
#pragma unroll
            for (int e = 0; e < 8; ++e) { g[i][e] = 0.f; xh[i][e] = 0.f; }
            if (c0 + 8 <= C) {
                short8v dyv = *reinterpret_cast<const short8v*>(dyrow + c0);
                short8v xv = *reinterpret_cast<const short8v*>(xrow + c0);
                short8v wv = *reinterpret_cast<const short8v*>(w + c0);
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    float d = bf2f((unsigned short)dyv[e]);
                    float gg = d * bf2f((unsigned short)wv[e]);
                    float xhat = (bf2f((unsigned short)xv[e]) - mean) * rstd;
                    g[i][e] = gg;
                    xh[i][e] = xhat;
                    s1 += gg;
                    s2 += gg * xhat;
                    dwacc[i][e] += d * xhat;
                    dbacc[i][e] += d;
                }
            } else if (c0 < C) {
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    if (c0 + e < C) {
                        float d = bf2f(dyrow[c0 + e]);
                        float gg = d * bf2f(w[c0 + e]);
                        float xhat = (bf2f(xrow[c0 + e]) - mean) * rstd;
                        g[i][e] = gg;
                        xh[i][e] = xhat;
                        s1 += gg;
                        s2 += gg * xhat;
                        dwacc[i][e] += d * xhat;
                        dbacc[i][e] += d;
                    }
                }
            }
        }
        s1 = wave_sum(s1) / C;
        s2 = wave_sum(s2) / C;
#pragma unroll
        for (int i = 0; i < CHUNKS; ++i) {
            int c0 = lane * 8 + i * 512;
            if (c0 + 8 <= C) {
                short8v o;
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    o[e] = (short)f2bf(rstd * (g[i][e] - s1 - xh[i][e] * s2));
                *reinterpret_cast<short8v*>(dxrow + c0) = o;
            } else if (c0 < C) {
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    if (c0 + e < C)
                        dxrow[c0 + e] = f2bf(rstd * (g[i][e] - s1 - xh[i][e] * s2));
            }
        }
    }

    // combine the block's waves (all cover the same column range) and emit
    // one tiled partial row (layout shared with the colsum finalize)
    __shared__ float red[2][2048];
    for (int i = threadIdx.x; i < 2048; i += blockDim.x) {
        red[0][i] = 0.f;
        red[1][i] = 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
        int c0 = lane * 8 + i * 512;
#pragma unroll
        for (int e = 0; e < 8; ++e)
            if (c0 + e < C) {
                atomicAdd(&red[0][c0 + e], dwacc[i][e]);  // LDS only
                atomicAdd(&red[1][c0 + e], dbacc[i][e]);
            }
    }
    __syncthreads();
    for (int c = threadIdx.x; c < C; c += blockDim.x) {
        partial[((long)(c >> 6) * gridDim.x + blockIdx.x) * DW_LANES + (c & 63)] =
            red[0][c];
        int c2 = C + c;
        partial[((long)(c2 >> 6) * gridDim.x + blockIdx.x) * DW_LANES + (c2 & 63)] =
            red[1][c];
    }
}

template <int CHUNKS>
void launch_ln_fwd(const torch::Tensor& x, const torch::Tensor& w, const unsigned short* bp,
                   torch::Tensor& y, torch::Tensor& mean, torch::Tensor& rstd,
                   long rows, int C, float eps) {
    int wpb = 4;
    long blocks = (rows + wpb - 1) / wpb;
    hipLaunchKernelGGL((ln_fwd_kernel<CHUNKS>), dim3(blocks), dim3(64 * wpb), 0,
                       at::cuda::getCurrentCUDAStream(),
                       reinterpret_cast<const unsigned short*>(x.data_ptr()),
                       reinterpret_cast<const unsigned short*>(w.data_ptr()), bp,
                       reinterpret_cast<unsigned short*>(y.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), rows, C, eps);
    HIP_CHECK_LAST();
}

template <int CHUNKS>
void launch_ln_bwd_dx(const torch::Tensor& dy, const torch::Tensor& x, const torch::Tensor& w,
                      const torch::Tensor& mean, const torch::Tensor& rstd, torch::Tensor& dx,
                      long rows, int C) {
    int wpb = 4;
    long blocks = (rows + wpb - 1) / wpb;
    hipLaunchKernelGGL((ln_bwd_dx_kernel<CHUNKS>), dim3(blocks), dim3(64 * wpb), 0,
                       at::cuda::getCurrentCUDAStream(),
                       reinterpret_cast<const unsigned short*>(dy.data_ptr()),
                       reinterpret_cast<const unsigned short*>(x.data_ptr()),
                       reinterpret_cast<const unsigned short*>(w.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(dx.data_ptr()), rows, C);
    HIP_CHECK_LAST();
}

template <int CHUNKS>
long launch_ln_bwd_fused(const torch::Tensor& dy, const torch::Tensor& x,
                         const torch::Tensor& w, const torch::Tensor& mean,
                         const torch::Tensor& rstd, torch::Tensor& dx,
                         torch::Tensor& partial, long rows, int C, long nblocks) {
    int wpb = 4;
    hipLaunchKernelGGL((ln_bwd_fused_kernel<CHUNKS>), dim3(nblocks), dim3(64 * wpb), 0,
                       at::cuda::getCurrentCUDAStream(),
                       reinterpret_cast<const unsigned short*>(dy.data_ptr()),
                       reinterpret_cast<const unsigned short*>(x.data_ptr()),
                       reinterpret_cast<const unsigned short*>(w.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(dx.data_ptr()),
                       partial.data_ptr<float>(), rows, C);
    HIP_CHECK_LAST();
    return nblocks;
}

}  // namespace

std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> b,
                                  double eps) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
    x = x.contiguous();
    auto wc = w.contiguous();
    int C = x.size(-1);
    TORCH_CHECK(C <= 2048, "ln_fwd: C must be <= 2048");
    long rows = x.numel() / C;
    auto y = torch::empty_like(x);
    auto mean = torch::empty({rows}, x.options().dtype(torch::kFloat32));
    auto rstd = torch::empty_like(mean);
    if (rows == 0) return {y, mean, rstd};
    const unsigned short* bp = nullptr;
    torch::Tensor bc;
    if (b.has_value() && b->defined()) {
        bc = b->contiguous();
        bp = reinterpret_cast<const unsigned short*>(bc.data_ptr());
    }
    int chunks = (C + 511) / 512;
    if (chunks == 1)      launch_ln_fwd<1>(x, wc, bp, y, mean, rstd, rows, C, (float)eps);
    else if (chunks == 2) launch_ln_fwd<2>(x, wc, bp, y, mean, rstd, rows, C, (float)eps);
    else if (chunks == 3) launch_ln_fwd<3>(x, wc, bp, y, mean, rstd, rows, C, (float)eps);
    else                  launch_ln_fwd<4>(x, wc, bp, y, mean, rstd, rows, C, (float)eps);
    return {y, mean, rstd};
}

std::vector<torch::Tensor> ln_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                  torch::Tensor mean, torch::Tensor rstd, bool needs_dwdb) {
    dy = dy.contiguous(); x = x.contiguous();
    auto wc = w.contiguous();
    int C = x.size(-1);
    long rows = x.numel() / C;
    auto dx = torch::empty_like(x);
    if (rows == 0) return {dx, torch::zeros_like(w), torch::zeros_like(w)};
    int chunks = (C + 511) / 512;
    if (!needs_dwdb) {
        if (chunks == 1)      launch_ln_bwd_dx<1>(dy, x, wc, mean, rstd, dx, rows, C);
        else if (chunks == 2) launch_ln_bwd_dx<2>(dy, x, wc, mean, rstd, dx, rows, C);
        else if (chunks == 3) launch_ln_bwd_dx<3>(dy, x, wc, mean, rstd, dx, rows, C);
        else                  launch_ln_bwd_dx<4>(dy, x, wc, mean, rstd, dx, rows, C);
        return {dx, torch::Tensor(), torch::Tensor()};
    }

    // fused single pass: dx + per-block dw/db partials + shared finalize
    static const long kRedBlocks = [] {
        const char* e = getenv("PERCEIVER_RED_BLOCKS");
        return e ? atol(e) : 1024L;
    }();
    // row-adaptive block count (same sweep-derived heuristic as colsum)
    long nblocks = std::min(std::max(rows / 32, (long)256), kRedBlocks);
    nblocks = std::min(nblocks, (rows + 3) / 4);
    long cgroups = (2 * (long)C + DW_LANES - 1) / DW_LANES;
    auto partial = torch::empty({cgroups * nblocks, (long)DW_LANES},
                                x.options().dtype(torch::kFloat32));
    if (chunks == 1)      launch_ln_bwd_fused<1>(dy, x, wc, mean, rstd, dx, partial, rows, C, nblocks);
    else if (chunks == 2) launch_ln_bwd_fused<2>(dy, x, wc, mean, rstd, dx, partial, rows, C, nblocks);
    else if (chunks == 3) launch_ln_bwd_fused<3>(dy, x, wc, mean, rstd, dx, partial, rows, C, nblocks);
    else                  launch_ln_bwd_fused<4>(dy, x, wc, mean, rstd, dx, partial, rows, C, nblocks);
    auto acc = torch::empty({2, (long)C}, x.options());  // bf16 straight out
    colsum_reduce_partials(partial, acc, (int)nblocks, 2 * C);
    return {dx, acc[0], acc[1]};
}
