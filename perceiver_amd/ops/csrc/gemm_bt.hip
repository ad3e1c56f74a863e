// Custom bf16 GEMM for the Perceiver projection shapes: y = x @ w^T (+ bias).
//   x: (M, K) bf16 row-major        w: (N, K) bf16 row-major (nn.Linear layout)
//   y: (M, N) bf16
//
// hipBLASLt (TunableOp-tuned) averages ~790 TF/s on the MLM flagship's
// projection/MLP shapes (M = 16384, K/N 1280-2816) — the largest single block
// of the training step. This kernel applies the CDNA4 deep-pipeline GEMM
// recipe from the gfx950 guide:
//   - 256x128 output tile, BK = 64, 8 waves (4M x 2N), 512 threads,
//     one block/CU (the 256x128 tile also fills the grid better than 256^2
//     on N = 1280: 640 workgroups vs 320);
//   - global_load_lds (16 B/lane) staging into a 3-deep LDS ring
//     (48 KB/buffer = 144 KB), two K-tiles of prefetch lead, COUNTED
//     s_waitcnt vmcnt — the main loop never drains to vmcnt(0), so HBM
//     traffic stays in flight across the raw s_barriers;
//   - a conflict-free XOR granule swizzle applied on the glds SOURCE
//     addresses and repeated on the ds_read offsets — glds placement is
//     forced lane-linear, so the swizzle has to ride on which global granule
//     each lane fetches (see swz() for the bank derivation);
//   - phase = [vmcnt][s_barrier][issue next tile's glds][ds_read 12 frags]
//     [setprio(1) 16x MFMA setprio(0)]; two phases per K-tile (mf halves);
//   - C staged through LDS after the loop (buffers are dead by then) and
//     stored as 16-B row chunks with the bias added in the epilogue, which
//     also replaces the separate bias elementwise pass.
//
// Both operands are K-contiguous (x rows and nn.Linear weight rows), so A and
// B stage through the identical path — the B^T orientation is what makes the
// deep pipeline this uniform, and it is the layout every Linear in the model
// already has.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

namespace {

constexpr int BM = 256;
constexpr int BN = 128;
constexpr int BK = 64;
constexpr int THREADS = 512;          // 8 waves: 4 (M) x 2 (N)
constexpr int WM = 64;                // wave tile 64 x 64
constexpr int WN = 64;
constexpr int A_BYTES = BM * BK * 2;  // 32 KB
constexpr int B_BYTES = BN * BK * 2;  // 16 KB
constexpr int BUF_BYTES = A_BYTES + B_BYTES;   // 48 KB; x3 ring = 144 KB
constexpr int GLDS_A = A_BYTES / (8 * 1024);   // glds per wave per tile (A): 4
constexpr int GLDS_B = B_BYTES / (8 * 1024);   // 2

// Granule swizzle matched to the fragment-read pattern. Reads fetch 16 B at
// byte (row*128 + c*16) with a quarter-wave covering rows r0..r0+15 at fixed
// c; the bank group of a 16-B granule is (byte>>4) & 15, which unswizzled
// depends on row parity only -> 8-way conflicts (PMC measured 2e8 conflict
// cycles; hipBLASLt has zero). XORing the granule-within-row bits (4..6)
// with (row>>1)&7 makes (row&1, c^((row>>1)&7)) injective over 16 rows: all
// 16 lanes of a quarter-wave hit distinct bank groups. Involution, moves
// whole granules, and stays inside each 128-B row.
DEVINL int swz(int byte_off) {
    return byte_off ^ (((byte_off >> 8) & 7) << 4);
}

// one wave-level glds: every lane deposits 16 B at lds_base + lane*16;
// the LDS operand must be wave-uniform
DEVINL void glds16(const unsigned short* gsrc_lane, char* lds_uniform) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gsrc_lane,
        (__attribute__((address_space(3))) unsigned int*)lds_uniform, 16, 0, 0);
}

// Issue the glds for one operand tile. The LDS image is lane-linear
// row-major [rows][BK]; the source granule for linear offset o is the one
// that belongs at swz(o), which inverts the swizzle on the read side.
DEVINL void stage_tile(const unsigned short* __restrict__ gbase, long ld,
                       char* lds_base, int wid, int lane, int n_glds) {
    for (int i = 0; i < n_glds; ++i) {
        int piece = wid * n_glds + i;
        int o = piece * 1024 + lane * 16;       // linear LDS byte offset
        int so = swz(o);
        int row = so / (BK * 2);
        int col2 = so % (BK * 2);
        glds16(gbase + (long)row * ld + col2 / 2, lds_base + piece * 1024);
    }
}

__launch_bounds__(THREADS, 1)
__global__ void gemm_bt_kernel(const unsigned short* __restrict__ xp,   // (M,K)
                               const unsigned short* __restrict__ wp,   // (N,K)
                               const unsigned short* __restrict__ bias, // (N) or null
                               unsigned short* __restrict__ yp,         // (M,N)
                               int M, int N, int K) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = __builtin_amdgcn_readfirstlane(tid >> 6);
    const int lo16 = lane & 15;
    const int hi4 = lane >> 4;

    // XCD-aware bijective block swizzle: contiguous grid chunks per XCD so
    // neighboring tiles (sharing A rows / B cols) hit the same L2
    int bid = blockIdx.x;
    {
        int nwg = gridDim.x;
        int q = nwg / 8, r = nwg % 8;
        int xcd = bid % 8, idx = bid / 8;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    const int n_tiles_n = N / BN;
    const int m0 = (bid / n_tiles_n) * BM;
    const int n0 = (bid % n_tiles_n) * BN;

    const int wm = wid >> 1;          // 0..3 -> 64-row slice of the tile
    const int wn = wid & 1;           // 0..1 -> 64-col slice
    const int arow0 = wm * WM;
    const int bcol0 = wn * WN;

    float4v acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = float4v{0.f, 0.f, 0.f, 0.f};

    const unsigned short* abase = xp + (long)m0 * K;
    const unsigned short* bbase = wp + (long)n0 * K;
    const int kt_total = K / BK;

    auto issue_tile = [&](int kt) {
        char* buf = smem + (kt % 3) * BUF_BYTES;
        stage_tile(abase + (long)kt * BK, K, buf, wid, lane, GLDS_A);
        stage_tile(bbase + (long)kt * BK, K, buf + A_BYTES, wid, lane, GLDS_B);
    };

    // fragment read from buffer `base`: operand row (within tile), k-block kk
    // (32 wide): elem (lo16-row, hi4*8+e) per the 16x16x32 A/B layout
    auto read_frag = [&](const char* base, int row, int kk) -> bf16x8 {
        int o = row * (BK * 2) + kk * 64 + hi4 * 16;
        return (bf16x8)(*reinterpret_cast<const short8v*>(base + swz(o)));
    };

    // ---- prologue: 2 K-tiles of lead (applicable() guarantees >= 3) ----
    issue_tile(0);
    issue_tile(1);

    // One phase: half the wave's C rows x full K-tile. VM in GEMM_SYNC is
    // the counted vmcnt: 6 = one tile (6 glds/wave) still in flight, 0 =
    // full drain (only for the final tiles). KT_NEXT < kt_total gates the
    // prefetch, which issues right after the sync barrier.
    // Sync happens ONCE per K-tile (before the half-0 phase): the vmcnt
    // forces this tile's glds landed, the barrier makes every wave's visible
    // AND proves the previous tile's readers are done before the prefetch
    // below overwrites its ring slot. The half-1 phase rides with no sync at
    // all: its tile was forced by its half-0 sibling, and the next half-0
    // barrier orders any later overwrite against its reads.
#define GEMM_SYNC(VM)                                                                  \
    asm volatile("s_waitcnt vmcnt(" #VM ")" ::: "memory");                             \
    __builtin_amdgcn_s_barrier();

    // Whole K-tile: the 4 B fragments are shared by both mf halves, so they
    // load once per tile (8 instead of 12 ds_read_b128 per phase). A-frags
    // for the second half load before the first half's MFMA cluster so the
    // LDS pipe work hides under the matrix pipe.
#define GEMM_TILE(KT, KT_NEXT)                                                         \
    {                                                                                  \
        if ((KT_NEXT) < kt_total) issue_tile(KT_NEXT);                                 \
        const char* abuf = smem + ((KT) % 3) * BUF_BYTES;                              \
        const char* bbuf = abuf + A_BYTES;                                             \
        bf16x8 afrag[4][2], bfrag[4][2];                                               \
        _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                               \
            _Pragma("unroll") for (int kk = 0; kk < 2; ++kk)                           \
                bfrag[nf][kk] = read_frag(bbuf, bcol0 + nf * 16 + lo16, kk);           \
        _Pragma("unroll") for (int mf = 0; mf < 4; ++mf)                               \
            _Pragma("unroll") for (int kk = 0; kk < 2; ++kk)                           \
                afrag[mf][kk] = read_frag(abuf, arow0 + mf * 16 + lo16, kk);           \
        __builtin_amdgcn_s_setprio(1);                                                 \
        /* kk outermost: the 8 accumulators between dependent kk pairs keep */         \
        /* the MFMA pipe free of read-after-write stalls */                            \
        _Pragma("unroll") for (int half = 0; half < 2; ++half)                         \
            _Pragma("unroll") for (int kk = 0; kk < 2; ++kk)                           \
                _Pragma("unroll") for (int mf = 0; mf < 2; ++mf)                       \
                    _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                   \
                        acc[half * 2 + mf][nf] =                                       \
                            __builtin_amdgcn_mfma_f32_16x16x32_bf16(                   \
                                afrag[half * 2 + mf][kk], bfrag[nf][kk],               \
                                acc[half * 2 + mf][nf], 0, 0, 0);                      \
        __builtin_amdgcn_s_setprio(0);                                                 \
    }

    // peel the last triple so the tail below always knows its drain counts
    const int kt_main = (kt_total / 3) * 3;
    const int kt_peel = (kt_total % 3 == 0) ? kt_main - 3 : kt_main;
    // MF_HALF==1 phases read the same tile their half-0 sibling already
    // forced, so their wait is vmcnt(12) = never stalls (outstanding is at
    // most two tiles); a 6 there was measured to serialize the prefetch lead.
    int kt = 0;
    for (; kt < kt_peel; kt += 3) {
        GEMM_SYNC(6)
        GEMM_TILE(kt + 0, kt + 2)
        GEMM_SYNC(6)
        GEMM_TILE(kt + 1, kt + 3)
        GEMM_SYNC(6)
        GEMM_TILE(kt + 2, kt + 4)
    }
    // tail: 1..3 tiles left, nothing further to prefetch past kt_total
    switch (kt_total - kt) {
        case 3:
            GEMM_SYNC(6)
            GEMM_TILE(kt + 0, kt + 2)
            GEMM_SYNC(6)
            GEMM_TILE(kt + 1, kt_total)
            GEMM_SYNC(0)
            GEMM_TILE(kt + 2, kt_total)
            break;
        case 2:
            GEMM_SYNC(6)
            GEMM_TILE(kt + 0, kt_total)
            GEMM_SYNC(0)
            GEMM_TILE(kt + 1, kt_total)
            break;
        default:
            GEMM_SYNC(0)
            GEMM_TILE(kt + 0, kt_total)
            break;
    }
#undef GEMM_TILE
#undef GEMM_SYNC

    // ---- epilogue: stage C through LDS (ring is dead), bias, 16-B stores ----
    // bias is added in fp32 BEFORE the bf16 round, matching hipBLASLt's
    // epilogue exactly (rounding C first then adding bias flips ~25% of
    // outputs by one ulp vs F.linear)
    float bvals[4] = {0.f, 0.f, 0.f, 0.f};
    if (bias != nullptr) {
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
            bvals[nf] = bf2f(bias[n0 + bcol0 + nf * 16 + lo16]);
    }
    __builtin_amdgcn_s_barrier();
    char* cmine = smem + wid * (WM * WN * 2);  // 8 KB per wave
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = mf * 16 + hi4 * 4 + r;
                int col = nf * 16 + lo16;
                *reinterpret_cast<unsigned short*>(cmine + (row * WN + col) * 2) =
                    f2bf(acc[mf][nf][r] + bvals[nf]);
            }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // own-wave LDS ordering
#pragma unroll
    for (int i = 0; i < (WM * WN * 2) / (64 * 16); ++i) {
        int g = lane + i * 64;
        int row = g / (WN / 8);
        int c0 = (g % (WN / 8)) * 8;
        short8v v = *reinterpret_cast<const short8v*>(cmine + (row * WN + c0) * 2);
        *reinterpret_cast<short8v*>(yp + ((long)m0 + arow0 + row) * N + n0 + bcol0 + c0) = v;
    }
}

}  // namespace

bool gemm_bt_applicable(long M, long N, long K) {
    return M % BM == 0 && N % BN == 0 && K % BK == 0 && K >= 3 * BK &&
           M > 0 && K <= (1 << 18);
}

torch::Tensor gemm_bt_bf16(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16, "gemm_bt: x must be cuda bf16");
    TORCH_CHECK(w.scalar_type() == torch::kBFloat16, "gemm_bt: w must be bf16");
    TORCH_CHECK(x.dim() == 2 && w.dim() == 2);
    x = x.contiguous();
    auto wc = w.contiguous();
    long M = x.size(0), K = x.size(1), N = wc.size(0);
    TORCH_CHECK(wc.size(1) == K, "gemm_bt: inner dims mismatch");
    TORCH_CHECK(gemm_bt_applicable(M, N, K), "gemm_bt: unsupported shape ", M, "x", N, "x", K);
    auto y = torch::empty({M, N}, x.options());
    const unsigned short* bp = nullptr;
    torch::Tensor bc;
    if (bias.has_value() && bias->defined()) {
        bc = bias->contiguous();
        TORCH_CHECK(bc.numel() == N && bc.scalar_type() == torch::kBFloat16);
        bp = reinterpret_cast<const unsigned short*>(bc.data_ptr());
    }
    static bool raised = [] {
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&gemm_bt_kernel),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        return true;
    }();
    (void)raised;
    long grid = (M / BM) * (N / BN);
    hipLaunchKernelGGL(gemm_bt_kernel, dim3(grid), dim3(THREADS), 3 * BUF_BYTES,
                       at::cuda::getCurrentCUDAStream(),
                       reinterpret_cast<const unsigned short*>(x.data_ptr()),
                       reinterpret_cast<const unsigned short*>(wc.data_ptr()),
                       bp, reinterpret_cast<unsigned short*>(y.data_ptr()),
                       (int)M, (int)N, (int)K);
    HIP_CHECK_LAST();
    return y;
}
