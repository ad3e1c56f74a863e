// Custom bf16 GEMM for the Perceiver projection shapes: y = x @ w^T (+ bias),
// with an optional fused-GELU second output for the MLP widening layer.
//   x: (M, K) bf16 row-major        w: (N, K) bf16 row-major (nn.Linear layout)
//   y: (M, N) bf16                  yg (optional): gelu(y + bias)
//
// hipBLASLt (TunableOp-tuned) runs ~800-1100 TF/s on the MLM flagship's
// projection/MLP shapes (M = 16384, K/N 768-2816) — the largest single block
// of the training step. This kernel applies the CDNA4 deep-pipeline GEMM
// recipe from the gfx950 guide:
//   - 256xBN output tile (BN 128 or 160), BK = 64, 8 waves (4M x 2N),
//     512 threads, one block/CU. BN=160 exists because the dominant N=1280
//     class then maps to grids that FILL the chip exactly (16384x1280:
//     512 workgroups = 2.0 rounds of 256 CUs; at BN=128 it is 640 = 2.5
//     rounds, i.e. 17% idle in the tail round — measured as the main gap
//     vs hipBLASLt's MT160x256 pick);
//   - global_load_lds staging into a 3-deep LDS ring, two K-tiles of
//     prefetch lead, COUNTED s_waitcnt vmcnt — the main loop never drains
//     to vmcnt(0), so HBM traffic stays in flight across the raw s_barriers
//     (one sync per K-tile);
//   - a conflict-free XOR granule swizzle applied on the glds SOURCE
//     addresses and repeated on the ds_read offsets — glds placement is
//     forced lane-linear, so the swizzle rides on which global granule each
//     lane fetches (see swz() for the bank derivation; PMC measured 2e8
//     LDS conflict cycles with the naive layout, ~0 after);
//   - B fragments load once per K-tile (shared by both mf halves);
//   - C staged through LDS after the loop (the ring is dead by then) and
//     stored as 16-B row chunks with the bias added in fp32 in the epilogue;
//     the GELU variant additionally writes gelu(pre + bias) so the MLP's
//     separate bias+GELU elementwise pass (and its extra activation read)
//     disappears while the saved pre-activation keeps gelu_bias_bwd's
//     contract unchanged.
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
constexpr int BK = 64;
constexpr int THREADS = 512;          // 8 waves: 4 (M) x 2 (N)
constexpr int WM = 64;                // wave rows
constexpr int A_BYTES = BM * BK * 2;  // 32 KB
constexpr int GLDS_A = A_BYTES / (8 * 1024);   // 16-B glds per wave (A): 4

// Granule swizzle matched to the fragment-read pattern. Reads fetch 16 B at
// byte (row*128 + c*16) with a quarter-wave covering rows r0..r0+15 at fixed
// c; the bank group of a 16-B granule is (byte>>4) & 15, which unswizzled
// depends on row parity only -> 8-way conflicts. XORing the
// granule-within-row bits (4..6) with (row>>1)&7 makes (row&1, c^((row>>1)&7))
// injective over 16 rows: all 16 lanes of a quarter-wave hit distinct bank
// groups. Involution, moves whole 16-B granules, stays inside each 128-B row.
DEVINL int swz(int byte_off) {
    return byte_off ^ (((byte_off >> 8) & 7) << 4);
}

// wave-level glds, 16 B/lane; LDS operand must be wave-uniform
DEVINL void glds16(const unsigned short* gsrc_lane, char* lds_uniform) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gsrc_lane,
        (__attribute__((address_space(3))) unsigned int*)lds_uniform, 16, 0, 0);
}

// wave-level glds, 4 B/lane (for tile sizes that do not split into uniform
// per-wave 1-KB pieces; uniformity keeps the per-wave vmcnt counts exact)
DEVINL void glds4(const unsigned short* gsrc_lane, char* lds_uniform) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gsrc_lane,
        (__attribute__((address_space(3))) unsigned int*)lds_uniform, 4, 0, 0);
}

// A-tile staging: 16-B lanes, 1-KB pieces, 4 per wave. The LDS image is
// lane-linear row-major [rows][BK]; the source granule for linear offset o
// is the one that belongs at swz(o).
DEVINL void stage_a(const unsigned short* __restrict__ gbase, long ld,
                    char* lds_base, int wid, int lane) {
#pragma unroll
    for (int i = 0; i < GLDS_A; ++i) {
        int piece = wid * GLDS_A + i;
        int o = piece * 1024 + lane * 16;
        int so = swz(o);
        glds16(gbase + (long)(so >> 7) * ld + ((so & 127) >> 1),
               lds_base + piece * 1024);
    }
}

// B-tile staging: 16-B lanes when the tile splits into uniform per-wave 1-KB
// pieces (BN=128: 2/wave — 4-B granules measured ~20% slower on the small-N
// shapes), else 4-B lanes in 256-B pieces (BN=160: 10/wave). Uniformity
// keeps the per-wave vmcnt counts exact.
template <int B_BYTES>
DEVINL void stage_b(const unsigned short* __restrict__ gbase, long ld,
                    char* lds_base, int wid, int lane) {
    if constexpr (B_BYTES % (8 * 1024) == 0) {
        constexpr int PER_WAVE = B_BYTES / (8 * 1024);
#pragma unroll
        for (int i = 0; i < PER_WAVE; ++i) {
            int piece = wid * PER_WAVE + i;
            int o = piece * 1024 + lane * 16;
            int so = swz(o);
            glds16(gbase + (long)(so >> 7) * ld + ((so & 127) >> 1),
                   lds_base + piece * 1024);
        }
    } else {
        constexpr int PER_WAVE = B_BYTES / 256 / 8;
#pragma unroll
        for (int i = 0; i < PER_WAVE; ++i) {
            int piece = wid * PER_WAVE + i;
            int o = piece * 256 + lane * 4;
            int so = swz(o);
            glds4(gbase + (long)(so >> 7) * ld + ((so & 127) >> 1),
                  lds_base + piece * 256);
        }
    }
}

template <int TBN, bool GELU>
__launch_bounds__(THREADS, 1)
__global__ void gemm_bt_kernel(const unsigned short* __restrict__ xp,   // (M,K)
                               const unsigned short* __restrict__ wp,   // (N,K)
                               const unsigned short* __restrict__ bias, // (N) or null
                               unsigned short* __restrict__ yp,         // (M,N)
                               unsigned short* __restrict__ yg,         // gelu out or null
                               int M, int N, int K) {
    constexpr int NF = TBN / 32;                  // B frags per wave: 4 or 5
    constexpr int WN = TBN / 2;                   // wave cols: 64 or 80
    constexpr int B_BYTES = TBN * BK * 2;         // 16 KB or 20 KB
    constexpr int BUF_BYTES = A_BYTES + B_BYTES;  // 48 KB or 52 KB
    // per-wave glds per K-tile: A 4 x 16 B + B (2 x 16 B, or 10 x 4 B)
    constexpr int VM_TILE =
        GLDS_A + (B_BYTES % (8 * 1024) == 0 ? B_BYTES / (8 * 1024) : B_BYTES / 256 / 8);

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
    const int n_tiles_n = N / TBN;
    const int m0 = (bid / n_tiles_n) * BM;
    const int n0 = (bid % n_tiles_n) * TBN;

    const int arow0 = (wid >> 1) * WM;
    const int bcol0 = (wid & 1) * WN;

    float4v acc[4][NF];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < NF; ++j) acc[i][j] = float4v{0.f, 0.f, 0.f, 0.f};

    const unsigned short* abase = xp + (long)m0 * K;
    const unsigned short* bbase = wp + (long)n0 * K;
    const int kt_total = K / BK;

    auto issue_tile = [&](int kt) {
        char* buf = smem + (kt % 3) * BUF_BYTES;
        stage_a(abase + (long)kt * BK, K, buf, wid, lane);
        stage_b<B_BYTES>(bbase + (long)kt * BK, K, buf + A_BYTES, wid, lane);
    };

    // fragment read: operand row (within tile), k-block kk (32 wide):
    // elem (lo16-row, hi4*8+e) per the 16x16x32 A/B layout
    auto read_frag = [&](const char* base, int row, int kk) -> bf16x8 {
        int o = row * (BK * 2) + kk * 64 + hi4 * 16;
        return (bf16x8)(*reinterpret_cast<const short8v*>(base + swz(o)));
    };

    // ---- prologue: 2 K-tiles of lead (applicable() guarantees >= 3) ----
    issue_tile(0);
    issue_tile(1);

    // Sync ONCE per K-tile: the counted vmcnt forces this tile's glds landed
    // (VM = VM_TILE leaves exactly one prefetched tile in flight; 0 only for
    // the final drains), the barrier makes every wave's visible AND proves
    // the previous tile's readers are done before the prefetch below
    // overwrites its ring slot.
#define GEMM_SYNC(VM)                                                                  \
    asm volatile("s_waitcnt vmcnt(%0)" ::"i"(VM) : "memory");                          \
    __builtin_amdgcn_s_barrier();

    // Whole K-tile. NF=4 reads all fragments up front and runs one 32-MFMA
    // cluster (measured fastest; fits in 224 VGPRs). NF=5 cannot hold both
    // kk fragment sets without spilling (29 VGPRs), so it splits by kk with
    // only one set live. Accumulators repeat at distance >= 4*NF, so the
    // matrix pipe sees no read-after-write stalls either way.
#define GEMM_TILE(KT, KT_NEXT)                                                         \
    {                                                                                  \
        if ((KT_NEXT) < kt_total) issue_tile(KT_NEXT);                                 \
        const char* abuf = smem + ((KT) % 3) * BUF_BYTES;                              \
        const char* bbuf = abuf + A_BYTES;                                             \
        if constexpr (NF == 4) {                                                       \
            bf16x8 afrag[4][2], bfrag[NF][2];                                          \
            _Pragma("unroll") for (int nf = 0; nf < NF; ++nf)                          \
                _Pragma("unroll") for (int kk = 0; kk < 2; ++kk)                       \
                    bfrag[nf][kk] = read_frag(bbuf, bcol0 + nf * 16 + lo16, kk);       \
            _Pragma("unroll") for (int mf = 0; mf < 4; ++mf)                           \
                _Pragma("unroll") for (int kk = 0; kk < 2; ++kk)                       \
                    afrag[mf][kk] = read_frag(abuf, arow0 + mf * 16 + lo16, kk);       \
            __builtin_amdgcn_s_setprio(1);                                             \
            _Pragma("unroll") for (int kk = 0; kk < 2; ++kk)                           \
                _Pragma("unroll") for (int mf = 0; mf < 4; ++mf)                       \
                    _Pragma("unroll") for (int nf = 0; nf < NF; ++nf)                  \
                        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(         \
                            afrag[mf][kk], bfrag[nf][kk], acc[mf][nf], 0, 0, 0);       \
            __builtin_amdgcn_s_setprio(0);                                             \
        } else {                                                                       \
            _Pragma("unroll") for (int kk = 0; kk < 2; ++kk) {                         \
                bf16x8 afrag[4], bfrag[NF];                                            \
                _Pragma("unroll") for (int nf = 0; nf < NF; ++nf)                      \
                    bfrag[nf] = read_frag(bbuf, bcol0 + nf * 16 + lo16, kk);           \
                _Pragma("unroll") for (int mf = 0; mf < 4; ++mf)                       \
                    afrag[mf] = read_frag(abuf, arow0 + mf * 16 + lo16, kk);           \
                __builtin_amdgcn_s_setprio(1);                                         \
                _Pragma("unroll") for (int mf = 0; mf < 4; ++mf)                       \
                    _Pragma("unroll") for (int nf = 0; nf < NF; ++nf)                  \
                        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(         \
                            afrag[mf], bfrag[nf], acc[mf][nf], 0, 0, 0);               \
                __builtin_amdgcn_s_setprio(0);                                         \
            }                                                                          \
        }                                                                              \
    }

    // peel the last triple so the tail below always knows its drain counts
    const int kt_main = (kt_total / 3) * 3;
    const int kt_peel = (kt_total % 3 == 0) ? kt_main - 3 : kt_main;
    int kt = 0;
    for (; kt < kt_peel; kt += 3) {
        GEMM_SYNC(VM_TILE)
        GEMM_TILE(kt + 0, kt + 2)
        GEMM_SYNC(VM_TILE)
        GEMM_TILE(kt + 1, kt + 3)
        GEMM_SYNC(VM_TILE)
        GEMM_TILE(kt + 2, kt + 4)
    }
    // tail: 1..3 tiles left, nothing further to prefetch past kt_total
    switch (kt_total - kt) {
        case 3:
            GEMM_SYNC(VM_TILE)
            GEMM_TILE(kt + 0, kt + 2)
            GEMM_SYNC(VM_TILE)
            GEMM_TILE(kt + 1, kt_total)
            GEMM_SYNC(0)
            GEMM_TILE(kt + 2, kt_total)
            break;
        case 2:
            GEMM_SYNC(VM_TILE)
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
    // epilogue exactly. The GELU variant stages the PRE-bias value (that is
    // the tensor the backward recomputes from) and derives gelu(pre + bias)
    // at store time.
    float bvals[NF];
#pragma unroll
    for (int nf = 0; nf < NF; ++nf)
        bvals[nf] = (!GELU && bias != nullptr) ? bf2f(bias[n0 + bcol0 + nf * 16 + lo16]) : 0.f;
    __builtin_amdgcn_s_barrier();
    char* cmine = smem + wid * (WM * WN * 2);  // 8 or 10 KB per wave
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < NF; ++nf)
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
        long out_off = ((long)m0 + arow0 + row) * N + n0 + bcol0 + c0;
        *reinterpret_cast<short8v*>(yp + out_off) = v;
        if (GELU) {
            short8v bv = bias != nullptr
                ? *reinterpret_cast<const short8v*>(bias + n0 + bcol0 + c0)
                : short8v{};
            short8v gv;
#pragma unroll
            for (int e = 0; e < 8; ++e)
                gv[e] = (short)f2bf(
                    gelu_f(bf2f((unsigned short)v[e]) + bf2f((unsigned short)bv[e])));
            *reinterpret_cast<short8v*>(yg + out_off) = gv;
        }
    }
}

// best fill on a 256-CU chip at one block/CU: fewer idle slots in the last
// round wins; tie goes to the wider tile (fewer blocks, more B reuse).
// PERCEIVER_GEMM_BN=128|160 forces a tile for A/B measurement.
int pick_bn(long M, long N) {
    static const int forced = [] {
        const char* e = getenv("PERCEIVER_GEMM_BN");
        return e ? atoi(e) : 0;
    }();
    if (forced == 128 || forced == 160) {
        if (N % forced == 0) return forced;
    }
    auto waste = [&](long bn) -> long {
        if (N % bn != 0) return 1L << 60;
        long g = (M / BM) * (N / bn);
        return ((g + 255) / 256) * 256 - g;
    };
    long w160 = waste(160), w128 = waste(128);
    if (w160 <= w128) return 160;
    return 128;
}

}  // namespace

bool gemm_bt_applicable(long M, long N, long K) {
    return M % BM == 0 && (N % 128 == 0 || N % 160 == 0) && K % BK == 0 &&
           K >= 3 * BK && M > 0 && K <= (1 << 18);
}

static std::vector<torch::Tensor> gemm_bt_run(torch::Tensor x, torch::Tensor w,
                                              c10::optional<torch::Tensor> bias,
                                              bool with_gelu) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16, "gemm_bt: x must be cuda bf16");
    TORCH_CHECK(w.scalar_type() == torch::kBFloat16, "gemm_bt: w must be bf16");
    TORCH_CHECK(x.dim() == 2 && w.dim() == 2);
    x = x.contiguous();
    auto wc = w.contiguous();
    long M = x.size(0), K = x.size(1), N = wc.size(0);
    TORCH_CHECK(wc.size(1) == K, "gemm_bt: inner dims mismatch");
    TORCH_CHECK(gemm_bt_applicable(M, N, K), "gemm_bt: unsupported shape ", M, "x", N, "x", K);
    auto y = torch::empty({M, N}, x.options());
    torch::Tensor ygel;
    unsigned short* ygp = nullptr;
    if (with_gelu) {
        ygel = torch::empty({M, N}, x.options());
        ygp = reinterpret_cast<unsigned short*>(ygel.data_ptr());
    }
    const unsigned short* bp = nullptr;
    torch::Tensor bc;
    if (bias.has_value() && bias->defined()) {
        bc = bias->contiguous();
        TORCH_CHECK(bc.numel() == N && bc.scalar_type() == torch::kBFloat16);
        bp = reinterpret_cast<const unsigned short*>(bc.data_ptr());
    }
    int bn = pick_bn(M, N);
    static bool raised = [] {
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&gemm_bt_kernel<128, false>),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&gemm_bt_kernel<128, true>),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&gemm_bt_kernel<160, false>),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&gemm_bt_kernel<160, true>),
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        return true;
    }();
    (void)raised;
    auto launch = [&](auto kern, int tbn) {
        long grid = (M / BM) * (N / tbn);
        size_t smem = 3 * (size_t)(A_BYTES + tbn * BK * 2);
        hipLaunchKernelGGL(kern, dim3(grid), dim3(THREADS), smem,
                           at::cuda::getCurrentCUDAStream(),
                           reinterpret_cast<const unsigned short*>(x.data_ptr()),
                           reinterpret_cast<const unsigned short*>(wc.data_ptr()),
                           bp, reinterpret_cast<unsigned short*>(y.data_ptr()),
                           ygp, (int)M, (int)N, (int)K);
    };
    if (bn == 160) {
        if (with_gelu) launch(gemm_bt_kernel<160, true>, 160);
        else launch(gemm_bt_kernel<160, false>, 160);
    } else {
        if (with_gelu) launch(gemm_bt_kernel<128, true>, 128);
        else launch(gemm_bt_kernel<128, false>, 128);
    }
    HIP_CHECK_LAST();
    if (with_gelu) return {y, ygel};
    return {y};
}

torch::Tensor gemm_bt_bf16(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias) {
    return gemm_bt_run(std::move(x), std::move(w), std::move(bias), false)[0];
}

std::vector<torch::Tensor> gemm_bt_gelu_bf16(torch::Tensor x, torch::Tensor w,
                                             c10::optional<torch::Tensor> bias) {
    return gemm_bt_run(std::move(x), std::move(w), std::move(bias), true);
}
