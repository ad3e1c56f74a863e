// Fused rotary position embedding for CDNA4.
//
// Replaces the host-side chain (cos/sin materialize, rotate-half stack,
// two muls, add, cat, dtype cast — ~6 elementwise kernels and several fp32
// intermediates per application, core/position.py:56-67) with ONE kernel:
//
//   out[..., 2i]   = t[2i]   * cos(f_i) - t[2i+1] * sin(f_i)
//   out[..., 2i+1] = t[2i+1] * cos(f_i) + t[2i]   * sin(f_i)     (i < rot/2)
//   out[..., c]    = t[c]                                        (c >= rot)
//
// f is the interleave-repeated frequency table (entries 2i and 2i+1 equal, so
// only f[2i] is read), fp32 as on the host path; t is bf16 with arbitrary
// batch/head/row strides (cached-KV strided views pass through uncopied).
// The backward is the same rotation with sin negated (orthogonal transform),
// so one kernel serves both directions via `neg_sin`.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

__global__ void rotary_kernel(
    const unsigned short* __restrict__ tp,  // (B,H,N,D) bf16, strided
    const float* __restrict__ fp,           // (FB,N,ROT) fp32, strided (FB in {1,B})
    unsigned short* __restrict__ op,        // (B,H,N,D) contiguous
    long tsb, long tsh, long tsn,
    long fsb, long fsn,
    int B, int H, int N, int D, int rot, int neg_sin) {
    // one thread per channel PAIR of one (b,h,n) row for the rotated span,
    // plus the pass-through tail handled by the same index space
    long total = (long)B * H * N * (D / 2 + (D & 1));
    long pairs_per_row = D / 2 + (D & 1);
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (long)gridDim.x * blockDim.x) {
        long row = i / pairs_per_row;
        int pr = (int)(i % pairs_per_row);
        int c = pr * 2;
        int n = (int)(row % N);
        long bh = row / N;
        int b = (int)(bh / H);
        int hh = (int)(bh % H);

        const unsigned short* trow = tp + (long)b * tsb + (long)hh * tsh + (long)n * tsn;
        unsigned short* orow = op + row * D;
        if (c + 1 < rot) {
            float e = bf2f(trow[c]);
            float o = bf2f(trow[c + 1]);
            float f = fp[(long)(fsb ? b : 0) * fsb + (long)n * fsn + c];
            float cs, sn;
            __sincosf(f, &sn, &cs);
            if (neg_sin) sn = -sn;
            orow[c] = f2bf(e * cs - o * sn);
            orow[c + 1] = f2bf(o * cs + e * sn);
        } else {
            // pass-through pair (possibly a lone tail channel)
            orow[c] = trow[c];
            if (c + 1 < D) orow[c + 1] = trow[c + 1];
        }
    }
}

}  // namespace

torch::Tensor rotary_apply(torch::Tensor t, torch::Tensor frq, int64_t rot, bool neg_sin) {
    TORCH_CHECK(t.is_cuda() && t.dim() == 4 && t.scalar_type() == torch::kBFloat16,
                "rotary_apply: bf16 (B,H,N,D) expected");
    TORCH_CHECK(frq.dim() == 3 && frq.scalar_type() == torch::kFloat32,
                "rotary_apply: fp32 (FB,N,rot) frequency table expected");
    TORCH_CHECK(rot % 2 == 0 && rot <= t.size(3), "rotary_apply: bad rotate_dim");
    TORCH_CHECK(frq.size(1) == t.size(2), "rotary_apply: table rows != seq rows");
    if (t.stride(3) != 1) t = t.contiguous();
    if (frq.stride(2) != 1) frq = frq.contiguous();

    int B = t.size(0), H = t.size(1), N = t.size(2), D = t.size(3);
    auto out = torch::empty({B, H, N, D}, t.options());
    if (t.numel() == 0) return out;
    long total = (long)B * H * N * ((D + 1) / 2);
    int threads = 256;
    long blocks = std::min((total + threads - 1) / threads, (long)16384);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(rotary_kernel, dim3((unsigned)blocks), dim3(threads), 0, stream,
                       reinterpret_cast<const unsigned short*>(t.data_ptr()),
                       frq.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(out.data_ptr()),
                       t.stride(0), t.stride(1), t.stride(2),
                       frq.size(0) > 1 ? frq.stride(0) : 0, frq.stride(1),
                       B, H, N, D, (int)rot, (int)neg_sin);
    HIP_CHECK_LAST();
    return out;
}
