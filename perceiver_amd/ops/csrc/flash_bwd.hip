// Flash attention backward for CDNA4 (gfx950): recompute-based, two kernels
// (atomic-free): dQ parallel over Q blocks, dK/dV parallel over KV blocks.
//
//   delta_i = sum_c dO[i][c] * O[i][c]
//   P       = exp(QK^T - lse_i)            (masks reapplied)
//   dS      = P * (dO V^T - delta_i)
//   dQ      = dS K          dK = dS^T Q          dV = P^T dO
//
// Layouts mirror flash_fwd.hip: 4 waves x 16 rows, KVBLK=32 tiles, LDS staging with
// +16 B row padding, mfma_f32_16x16x32_bf16, C-layout -> A-layout redistribution of
// P/dS through per-wave LDS. q arrives pre-scaled so no extra scale appears here.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <cfloat>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

namespace {

constexpr int KVBLK = 32;
constexpr int QROWS = 16;
constexpr int NWAVES = 4;
constexpr int QBLK = QROWS * NWAVES;

// ---------------------------------------------------------------- delta kernel
__global__ void delta_kernel(const unsigned short* __restrict__ dout,
                             const unsigned short* __restrict__ out,
                             float* __restrict__ delta, long rows, int dv) {
    // one 64-lane wave per row chunk: thread covers one row with a strided loop
    long row = blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    int lane = threadIdx.x % 64;
    if (row >= rows) return;
    const unsigned short* a = dout + row * dv;
    const unsigned short* b = out + row * dv;
    float acc = 0.f;
    for (int c = lane * 2; c + 1 < dv; c += 128) {
        acc += bf2f(a[c]) * bf2f(b[c]) + bf2f(a[c + 1]) * bf2f(b[c + 1]);
    }
    if (dv % 2 == 1 && lane == 0) acc += bf2f(a[dv - 1]) * bf2f(b[dv - 1]);
#pragma unroll
    for (int m = 1; m < 64; m <<= 1) acc += __shfl_xor(acc, m, 64);
    if (lane == 0) delta[row] = acc;
}

DEVINL float warp16_sum(float x) {
#pragma unroll
    for (int m = 1; m < 16; m <<= 1) x += __shfl_xor(x, m, 64);
    return x;
}

// stage (rows x d) tile row-major into LDS (row stride ldst_bytes), zero-pad
DEVINL void stage_rm(const unsigned short* __restrict__ src, long src_stride,
                     int rows_valid, int rows_tile, int d, int d_pad,
                     char* lds, int ldst_bytes, int tid, int nthreads) {
    const int gpr = d_pad / 8;
    const int total = rows_tile * gpr;
    for (int g = tid; g < total; g += nthreads) {
        int row = g / gpr;
        int c0 = (g % gpr) * 8;
        short8v val = {};
        if (row < rows_valid) {
            if (c0 + 8 <= d) {
                val = *reinterpret_cast<const short8v*>(src + (long)row * src_stride + c0);
            } else {
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    val[e] = (c0 + e < d) ? (short)src[(long)row * src_stride + c0 + e] : (short)0;
            }
        }
        *reinterpret_cast<short8v*>(lds + row * ldst_bytes + c0 * 2) = val;
    }
}

// stage (rows x d) tile TRANSPOSED into LDS: lds row = channel (d_pad rows),
// col = source row (rows_tile cols, stride ldst_bytes = rows_tile*2+16)
DEVINL void stage_tr(const unsigned short* __restrict__ src, long src_stride,
                     int rows_valid, int rows_tile, int d, int d_pad,
                     char* ldsT, int ldst_bytes, int tid, int nthreads) {
    const int gpr = d_pad / 8;
    const int total = rows_tile * gpr;
    for (int g = tid; g < total; g += nthreads) {
        int row = g / gpr;
        int c0 = (g % gpr) * 8;
        short8v val = {};
        if (row < rows_valid && c0 < d) {
            if (c0 + 8 <= d) {
                val = *reinterpret_cast<const short8v*>(src + (long)row * src_stride + c0);
            } else {
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    val[e] = (c0 + e < d) ? (short)src[(long)row * src_stride + c0 + e] : (short)0;
            }
        }
#pragma unroll
        for (int e = 0; e < 8; ++e)
            *reinterpret_cast<unsigned short*>(ldsT + (c0 + e) * ldst_bytes + row * 2) =
                (unsigned short)val[e];
    }
}

// ---------------------------------------------------------------- dQ kernel
// grid.x over Q blocks, grid.y = B*H. Stages per KV tile: K row-major (for S),
// K^T (for dQ = dS K), V row-major (for dP = dO V^T).
template <int DMAX, int DVMAX>
__launch_bounds__(256)
__global__ void flash_dq_kernel(
    const unsigned short* __restrict__ qp, const unsigned short* __restrict__ kp,
    const unsigned short* __restrict__ vp, const unsigned short* __restrict__ dop,
    const float* __restrict__ lsep, const float* __restrict__ deltap,
    const bool* __restrict__ pad,
    unsigned short* __restrict__ dqp,
    int B, int H, int Nq, int Lk, int D, int Dv, int causal,
    float drop_p, unsigned long long drop_seed) {
    const int d_pad = (D + 31) & ~31;
    const int dv_pad = (Dv + 31) & ~31;
    const int d_blocks = d_pad / 32;
    const int dv_blocks32 = dv_pad / 32;

    const int tid = threadIdx.x, wave = tid / 64, lane = tid % 64;
    const int lo16 = lane & 15, hi4 = lane >> 4;
    const int bh = blockIdx.y, b = bh / H;
    const int q0 = blockIdx.x * QBLK + wave * QROWS;

    const unsigned short* qbase = qp + (long)bh * Nq * D;
    const unsigned short* kbase = kp + (long)bh * Lk * D;
    const unsigned short* vbase = vp + (long)bh * Lk * Dv;
    const unsigned short* dobase = dop + (long)bh * Nq * Dv;
    const float* lse_row = lsep + (long)bh * Nq;
    const float* delta_row = deltap + (long)bh * Nq;
    const bool* padrow = pad ? pad + (long)b * Lk : nullptr;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int k_stride = d_pad * 2 + 16;
    const int kt_stride = KVBLK * 2 + 16;   // K^T: d_pad rows x 32 keys
    const int v_stride = dv_pad * 2 + 16;
    char* k_lds = smem;                                  // KVBLK * k_stride
    char* kt_lds = k_lds + KVBLK * k_stride;             // DMAX * kt_stride
    char* v_lds = kt_lds + DMAX * kt_stride;             // KVBLK * v_stride
    char* p_lds = v_lds + KVBLK * v_stride;              // NWAVES * QROWS * kt_stride
    char* p_mine = p_lds + wave * QROWS * kt_stride;

    // Q and dO fragments (A layout: lane = row lo16, k = hi4*8+e)
    short8v q_frag[DMAX / 32];
    short8v do_frag[DVMAX / 32];
    float lse_r[4], delta_r[4];
    {
        int qi = q0 + lo16;
        bool valid = qi < Nq;
        int qc = valid ? qi : Nq - 1;
        const unsigned short* qrow = qbase + (long)qc * D;
        const unsigned short* dorow = dobase + (long)qc * Dv;
#pragma unroll
        for (int kb = 0; kb < DMAX / 32; ++kb) {
            short8v val = {};
            if (kb < d_blocks && valid) {
                int c0 = kb * 32 + hi4 * 8;
                if (c0 + 8 <= D) val = *reinterpret_cast<const short8v*>(qrow + c0);
                else {
#pragma unroll
                    for (int e = 0; e < 8; ++e) val[e] = (c0 + e < D) ? (short)qrow[c0 + e] : (short)0;
                }
            }
            q_frag[kb] = val;
        }
#pragma unroll
        for (int kb = 0; kb < DVMAX / 32; ++kb) {
            short8v val = {};
            if (kb < dv_blocks32 && valid) {
                int c0 = kb * 32 + hi4 * 8;
                if (c0 + 8 <= Dv) val = *reinterpret_cast<const short8v*>(dorow + c0);
                else {
#pragma unroll
                    for (int e = 0; e < 8; ++e) val[e] = (c0 + e < Dv) ? (short)dorow[c0 + e] : (short)0;
                }
            }
            do_frag[kb] = val;
        }
        // per C-layout rows
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int qi2 = q0 + hi4 * 4 + r;
            lse_r[r] = (qi2 < Nq) ? lse_row[qi2] : 0.f;
            delta_r[r] = (qi2 < Nq) ? delta_row[qi2] : 0.f;
        }
    }

    float4v dq_acc[DMAX / 16];
#pragma unroll
    for (int cb = 0; cb < DMAX / 16; ++cb) dq_acc[cb] = float4v{0.f, 0.f, 0.f, 0.f};

    int kv_end = Lk;
    if (causal) kv_end = min(Lk, Lk - Nq + blockIdx.x * QBLK + QBLK);

    for (int kv0 = 0; kv0 < kv_end; kv0 += KVBLK) {
        int rows_valid = min(KVBLK, Lk - kv0);
        __syncthreads();
        stage_rm(kbase + (long)kv0 * D, D, rows_valid, KVBLK, D, d_pad, k_lds, k_stride, tid, 256);
        stage_tr(kbase + (long)kv0 * D, D, rows_valid, KVBLK, D, d_pad, kt_lds, kt_stride, tid, 256);
        stage_rm(vbase + (long)kv0 * Dv, Dv, rows_valid, KVBLK, Dv, dv_pad, v_lds, v_stride, tid, 256);
        __syncthreads();

        // S = Q K^T (16 x 32)
        float4v s_acc[2] = {float4v{0.f, 0.f, 0.f, 0.f}, float4v{0.f, 0.f, 0.f, 0.f}};
#pragma unroll
        for (int kb = 0; kb < DMAX / 32; ++kb) {
            if (kb < d_blocks) {
#pragma unroll
                for (int keyblk = 0; keyblk < 2; ++keyblk) {
                    const char* src = k_lds + (keyblk * 16 + lo16) * k_stride + (kb * 32 + hi4 * 8) * 2;
                    bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
                    s_acc[keyblk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        (bf16x8)q_frag[kb], bfrag, s_acc[keyblk], 0, 0, 0);
                }
            }
        }

        // dP = dO V^T (16 x 32)
        float4v dp_acc[2] = {float4v{0.f, 0.f, 0.f, 0.f}, float4v{0.f, 0.f, 0.f, 0.f}};
#pragma unroll
        for (int kb = 0; kb < DVMAX / 32; ++kb) {
            if (kb < dv_blocks32) {
#pragma unroll
                for (int keyblk = 0; keyblk < 2; ++keyblk) {
                    const char* src = v_lds + (keyblk * 16 + lo16) * v_stride + (kb * 32 + hi4 * 8) * 2;
                    bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
                    dp_acc[keyblk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        (bf16x8)do_frag[kb], bfrag, dp_acc[keyblk], 0, 0, 0);
                }
            }
        }

        // dS = P * (dP - delta), P = exp(S - lse)  [C layout]
#pragma unroll
        for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int qi = q0 + hi4 * 4 + r;
                int j = kv0 + kb * 16 + lo16;
                bool masked = j >= Lk || (padrow && j < Lk && padrow[j]) ||
                              (causal && j > Lk - Nq + qi);
                float p = masked ? 0.f : expf(s_acc[kb][r] - lse_r[r]);
                float dprobs = dp_acc[kb][r];
                if (drop_p > 0.f) {
                    unsigned int thresh = (unsigned int)(drop_p * 4294967296.0);
                    bool kept = rng_hash(drop_seed, bh, qi, j) >= thresh;
                    dprobs = kept ? dprobs / (1.0f - drop_p) : 0.f;
                }
                float ds = p * (dprobs - delta_r[r]);
                // store dS to per-wave LDS for A-layout reload (bf16)
                *reinterpret_cast<unsigned short*>(p_mine + (hi4 * 4 + r) * kt_stride + (kb * 16 + lo16) * 2) =
                    f2bf(ds);
            }
        }
        __builtin_amdgcn_s_waitcnt(0);
        bf16x8 ds_frag = (bf16x8)(*reinterpret_cast<const short8v*>(
            p_mine + lo16 * kt_stride + hi4 * 8 * 2));

        // dQ += dS K : B[key k][col d] = K^T_lds[d][k]
#pragma unroll
        for (int cb = 0; cb < DMAX / 16; ++cb) {
            if (cb * 16 < d_pad) {
                const char* src = kt_lds + (cb * 16 + lo16) * kt_stride + hi4 * 8 * 2;
                bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
                dq_acc[cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ds_frag, bfrag, dq_acc[cb], 0, 0, 0);
            }
        }
    }

    // store dQ (C layout rows hi4*4+r, col lo16+16cb)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int qi = q0 + hi4 * 4 + r;
        if (qi >= Nq) continue;
        unsigned short* dqrow = dqp + ((long)bh * Nq + qi) * D;
#pragma unroll
        for (int cb = 0; cb < DMAX / 16; ++cb) {
            int c = cb * 16 + lo16;
            if (c < D) dqrow[c] = f2bf(dq_acc[cb][r]);
        }
    }
}

// ---------------------------------------------------------------- dK/dV kernel
// grid.x over KV blocks (64 keys per workgroup, 16 per wave), grid.y = B*H.
// Loops over Q tiles of 32 rows; stages Q row-major + Q^T, dO row-major + dO^T.
template <int DMAX, int DVMAX>
__launch_bounds__(256)
__global__ void flash_dkv_kernel(
    const unsigned short* __restrict__ qp, const unsigned short* __restrict__ kp,
    const unsigned short* __restrict__ vp, const unsigned short* __restrict__ dop,
    const float* __restrict__ lsep, const float* __restrict__ deltap,
    const bool* __restrict__ pad,
    unsigned short* __restrict__ dkp, unsigned short* __restrict__ dvp,
    int B, int H, int Nq, int Lk, int D, int Dv, int causal,
    float drop_p, unsigned long long drop_seed) {
    constexpr int QTILE = 32;
    const int d_pad = (D + 31) & ~31;
    const int dv_pad = (Dv + 31) & ~31;
    const int d_blocks = d_pad / 32;
    const int dv_blocks32 = dv_pad / 32;

    const int tid = threadIdx.x, wave = tid / 64, lane = tid % 64;
    const int lo16 = lane & 15, hi4 = lane >> 4;
    const int bh = blockIdx.y, b = bh / H;
    const int k0 = blockIdx.x * QBLK + wave * QROWS;  // this wave's first key row

    const unsigned short* qbase = qp + (long)bh * Nq * D;
    const unsigned short* kbase = kp + (long)bh * Lk * D;
    const unsigned short* vbase = vp + (long)bh * Lk * Dv;
    const unsigned short* dobase = dop + (long)bh * Nq * Dv;
    const float* lse_row = lsep + (long)bh * Nq;
    const float* delta_row = deltap + (long)bh * Nq;
    const bool* padrow = pad ? pad + (long)b * Lk : nullptr;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int q_stride = d_pad * 2 + 16;      // Q row-major: QTILE rows
    const int qt_stride = QTILE * 2 + 16;     // Q^T: d_pad rows
    const int do_stride = dv_pad * 2 + 16;    // dO row-major: QTILE rows
    const int dot_stride = QTILE * 2 + 16;    // dO^T: dv_pad rows
    char* q_lds = smem;
    char* qt_lds = q_lds + QTILE * q_stride;
    char* do_lds = qt_lds + DMAX * qt_stride;
    char* dot_lds = do_lds + QTILE * do_stride;
    char* p_lds = dot_lds + DVMAX * dot_stride;   // NWAVES * QROWS * qt_stride
    char* p_mine = p_lds + wave * QROWS * qt_stride;

    // K and V fragments (A layout: lane = key row lo16, k = ch hi4*8+e)
    short8v k_frag[DMAX / 32];
    short8v v_frag[DVMAX / 32];
    bool key_pad[4];
    {
        int ki = k0 + lo16;
        bool valid = ki < Lk;
        int kc = valid ? ki : Lk - 1;
        const unsigned short* krow = kbase + (long)kc * D;
        const unsigned short* vrow = vbase + (long)kc * Dv;
#pragma unroll
        for (int kb = 0; kb < DMAX / 32; ++kb) {
            short8v val = {};
            if (kb < d_blocks && valid) {
                int c0 = kb * 32 + hi4 * 8;
                if (c0 + 8 <= D) val = *reinterpret_cast<const short8v*>(krow + c0);
                else {
#pragma unroll
                    for (int e = 0; e < 8; ++e) val[e] = (c0 + e < D) ? (short)krow[c0 + e] : (short)0;
                }
            }
            k_frag[kb] = val;
        }
#pragma unroll
        for (int kb = 0; kb < DVMAX / 32; ++kb) {
            short8v val = {};
            if (kb < dv_blocks32 && valid) {
                int c0 = kb * 32 + hi4 * 8;
                if (c0 + 8 <= Dv) val = *reinterpret_cast<const short8v*>(vrow + c0);
                else {
#pragma unroll
                    for (int e = 0; e < 8; ++e) val[e] = (c0 + e < Dv) ? (short)vrow[c0 + e] : (short)0;
                }
            }
            v_frag[kb] = val;
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int ki2 = k0 + hi4 * 4 + r;
            key_pad[r] = (ki2 >= Lk) || (padrow && padrow[min(ki2, Lk - 1)]);
        }
    }

    float4v dk_acc[DMAX / 16];
    float4v dv_acc[DVMAX / 16];
#pragma unroll
    for (int cb = 0; cb < DMAX / 16; ++cb) dk_acc[cb] = float4v{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int cb = 0; cb < DVMAX / 16; ++cb) dv_acc[cb] = float4v{0.f, 0.f, 0.f, 0.f};

    // causal: key j interacts with q rows i >= j - (Lk - Nq); start q tile there
    int q_start = 0;
    if (causal) {
        int j_lo = blockIdx.x * QBLK;                 // workgroup-min key
        q_start = max(0, j_lo - (Lk - Nq));
        q_start = (q_start / QTILE) * QTILE;
    }

    for (int qt0 = q_start; qt0 < Nq; qt0 += QTILE) {
        int rows_valid = min(QTILE, Nq - qt0);
        __syncthreads();
        stage_rm(qbase + (long)qt0 * D, D, rows_valid, QTILE, D, d_pad, q_lds, q_stride, tid, 256);
        stage_tr(qbase + (long)qt0 * D, D, rows_valid, QTILE, D, d_pad, qt_lds, qt_stride, tid, 256);
        stage_rm(dobase + (long)qt0 * Dv, Dv, rows_valid, QTILE, Dv, dv_pad, do_lds, do_stride, tid, 256);
        stage_tr(dobase + (long)qt0 * Dv, Dv, rows_valid, QTILE, Dv, dv_pad, dot_lds, dot_stride, tid, 256);
        __syncthreads();

        // S^T = K Q^T (16 keys x 32 qrows): B[ch k][col i] = Q^T_lds[ch][i]
        float4v st_acc[2] = {float4v{0.f, 0.f, 0.f, 0.f}, float4v{0.f, 0.f, 0.f, 0.f}};
#pragma unroll
        for (int kb = 0; kb < DMAX / 32; ++kb) {
            if (kb < d_blocks) {
#pragma unroll
                for (int qb = 0; qb < 2; ++qb) {
                    // B fragment: B[k][i] where k = ch, i = q row; read Q^T row (ch) ... but
                    // fragment wants lane col = i (q row), k = hi4*8+e (ch): element = Q[i][ch]
                    // = qt_lds[ch][i] -> lane reads COLUMN of qt_lds. Instead read from q_lds:
                    // q_lds[i][ch] with i = qb*16+lo16, ch = kb*32+hi4*8.. contiguous. B[k][j]
                    // wants contiguous k per lane -> that's qt_lds[...]. We need B[k=ch][j=qrow]:
                    // lane lo16 = qrow j, elements e over ch: qt wrong; q_lds row j gives Q[j][ch]
                    // contiguous in ch: exactly B[k][j] elements for fixed j. Use q_lds.
                    const char* src = q_lds + (qb * 16 + lo16) * q_stride + (kb * 32 + hi4 * 8) * 2;
                    bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
                    st_acc[qb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        (bf16x8)k_frag[kb], bfrag, st_acc[qb], 0, 0, 0);
                }
            }
        }

        // dP^T = V dO^T (16 keys x 32 qrows): B[ch][qrow] from do_lds rows
        float4v dpt_acc[2] = {float4v{0.f, 0.f, 0.f, 0.f}, float4v{0.f, 0.f, 0.f, 0.f}};
#pragma unroll
        for (int kb = 0; kb < DVMAX / 32; ++kb) {
            if (kb < dv_blocks32) {
#pragma unroll
                for (int qb = 0; qb < 2; ++qb) {
                    const char* src = do_lds + (qb * 16 + lo16) * do_stride + (kb * 32 + hi4 * 8) * 2;
                    bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
                    dpt_acc[qb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        (bf16x8)v_frag[kb], bfrag, dpt_acc[qb], 0, 0, 0);
                }
            }
        }

        // P^T and dS^T in C layout: row = key hi4*4+r (global k0+...), col = qrow qb*16+lo16
        // write both to per-wave LDS (P^T for dV, dS^T for dK), A-layout reload
#pragma unroll
        for (int qb = 0; qb < 2; ++qb) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int ki = k0 + hi4 * 4 + r;
                int qi = qt0 + qb * 16 + lo16;
                bool masked = key_pad[r] || qi >= Nq || (causal && ki > Lk - Nq + qi);
                float lse_i = (qi < Nq) ? lse_row[qi] : 0.f;
                float delta_i = (qi < Nq) ? delta_row[qi] : 0.f;
                float p = masked ? 0.f : expf(st_acc[qb][r] - lse_i);
                float p_eff = p;         // probs actually used in the forward PV
                float dprobs = dpt_acc[qb][r];
                if (drop_p > 0.f) {
                    unsigned int thresh = (unsigned int)(drop_p * 4294967296.0);
                    bool kept = rng_hash(drop_seed, bh, qi, ki) >= thresh;
                    float inv_keep = 1.0f / (1.0f - drop_p);
                    p_eff = kept ? p * inv_keep : 0.f;
                    dprobs = kept ? dprobs * inv_keep : 0.f;
                }
                float ds = p * (dprobs - delta_i);
                char* slot = p_mine + (hi4 * 4 + r) * qt_stride + (qb * 16 + lo16) * 2;
                // one buffer used twice: first P^T (for dV), then dS^T (for dK)
                *reinterpret_cast<unsigned short*>(slot) = f2bf(p_eff);
                // stash ds in registers for the second pass
                st_acc[qb][r] = ds;  // reuse st_acc as ds storage
            }
        }
        __builtin_amdgcn_s_waitcnt(0);
        bf16x8 pt_frag = (bf16x8)(*reinterpret_cast<const short8v*>(
            p_mine + lo16 * qt_stride + hi4 * 8 * 2));

        // dV += P^T dO : B[qrow i][ch c] = do_lds... B[k=i][j=c]: lane j=c col, k=i:
        // element = dO[i][c] = dot_lds[c][i] contiguous in i. Use dO^T.
#pragma unroll
        for (int cb = 0; cb < DVMAX / 16; ++cb) {
            if (cb * 16 < dv_pad) {
                const char* src = dot_lds + (cb * 16 + lo16) * dot_stride + hi4 * 8 * 2;
                bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
                dv_acc[cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pt_frag, bfrag, dv_acc[cb], 0, 0, 0);
            }
        }

        // second pass: dS^T through LDS
        __builtin_amdgcn_s_waitcnt(0);
        __syncthreads();  // ensure all waves finished reading P before overwrite (same buffer, wave-local actually)
#pragma unroll
        for (int qb = 0; qb < 2; ++qb) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                *reinterpret_cast<unsigned short*>(
                    p_mine + (hi4 * 4 + r) * qt_stride + (qb * 16 + lo16) * 2) = f2bf(st_acc[qb][r]);
            }
        }
        __builtin_amdgcn_s_waitcnt(0);
        bf16x8 dst_frag = (bf16x8)(*reinterpret_cast<const short8v*>(
            p_mine + lo16 * qt_stride + hi4 * 8 * 2));

        // dK += dS^T Q : B[qrow i][ch d] = Q[i][d] = qt_lds[d][i]... need contiguous k=i:
        // qt_lds row d holds Q[.][d] over i contiguous -> use qt_lds.
#pragma unroll
        for (int cb = 0; cb < DMAX / 16; ++cb) {
            if (cb * 16 < d_pad) {
                const char* src = qt_lds + (cb * 16 + lo16) * qt_stride + hi4 * 8 * 2;
                bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
                dk_acc[cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dst_frag, bfrag, dk_acc[cb], 0, 0, 0);
            }
        }
    }

    // store dK/dV (C layout: key row hi4*4+r, col lo16+16cb)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int ki = k0 + hi4 * 4 + r;
        if (ki >= Lk) continue;
        unsigned short* dkrow = dkp + ((long)bh * Lk + ki) * D;
        unsigned short* dvrow = dvp + ((long)bh * Lk + ki) * Dv;
#pragma unroll
        for (int cb = 0; cb < DMAX / 16; ++cb) {
            int c = cb * 16 + lo16;
            if (c < D) dkrow[c] = f2bf(dk_acc[cb][r]);
        }
#pragma unroll
        for (int cb = 0; cb < DVMAX / 16; ++cb) {
            int c = cb * 16 + lo16;
            if (c < Dv) dvrow[c] = f2bf(dv_acc[cb][r]);
        }
    }
}

template <int DMAX, int DVMAX>
void launch_flash_bwd(const torch::Tensor& dout, const torch::Tensor& q, const torch::Tensor& k,
                      const torch::Tensor& v, const torch::Tensor& lse, const torch::Tensor& delta,
                      const c10::optional<torch::Tensor>& pad_mask, bool causal,
                      float drop_p, unsigned long long drop_seed,
                      torch::Tensor& dq, torch::Tensor& dk, torch::Tensor& dv) {
    int B = q.size(0), H = q.size(1), Nq = q.size(2), D = q.size(3);
    int Lk = k.size(2), Dv = v.size(3);
    const int d_pad = (D + 31) & ~31;
    const int dv_pad = (Dv + 31) & ~31;
    auto stream = at::cuda::getCurrentCUDAStream();
    const bool* padp = nullptr;
    if (pad_mask.has_value() && pad_mask->defined()) padp = pad_mask->data_ptr<bool>();

    {   // dQ
        int k_stride = d_pad * 2 + 16, kt_stride = KVBLK * 2 + 16, v_stride = dv_pad * 2 + 16;
        size_t smem = (size_t)KVBLK * k_stride + (size_t)DMAX * kt_stride +
                      (size_t)KVBLK * v_stride + (size_t)NWAVES * QROWS * kt_stride;
        dim3 grid((Nq + QBLK - 1) / QBLK, B * H);
        hipLaunchKernelGGL((flash_dq_kernel<DMAX, DVMAX>), grid, dim3(256), smem, stream,
                           reinterpret_cast<const unsigned short*>(q.data_ptr()),
                           reinterpret_cast<const unsigned short*>(k.data_ptr()),
                           reinterpret_cast<const unsigned short*>(v.data_ptr()),
                           reinterpret_cast<const unsigned short*>(dout.data_ptr()),
                           lse.data_ptr<float>(), delta.data_ptr<float>(), padp,
                           reinterpret_cast<unsigned short*>(dq.data_ptr()),
                           B, H, Nq, Lk, D, Dv, (int)causal, drop_p, drop_seed);
        HIP_CHECK_LAST();
    }
    {   // dK/dV
        constexpr int QTILE = 32;
        int q_stride = d_pad * 2 + 16, qt_stride = QTILE * 2 + 16;
        int do_stride = dv_pad * 2 + 16, dot_stride = QTILE * 2 + 16;
        size_t smem = (size_t)QTILE * q_stride + (size_t)DMAX * qt_stride +
                      (size_t)QTILE * do_stride + (size_t)DVMAX * dot_stride +
                      (size_t)NWAVES * QROWS * qt_stride;
        dim3 grid((Lk + QBLK - 1) / QBLK, B * H);
        hipLaunchKernelGGL((flash_dkv_kernel<DMAX, DVMAX>), grid, dim3(256), smem, stream,
                           reinterpret_cast<const unsigned short*>(q.data_ptr()),
                           reinterpret_cast<const unsigned short*>(k.data_ptr()),
                           reinterpret_cast<const unsigned short*>(v.data_ptr()),
                           reinterpret_cast<const unsigned short*>(dout.data_ptr()),
                           lse.data_ptr<float>(), delta.data_ptr<float>(), padp,
                           reinterpret_cast<unsigned short*>(dk.data_ptr()),
                           reinterpret_cast<unsigned short*>(dv.data_ptr()),
                           B, H, Nq, Lk, D, Dv, (int)causal, drop_p, drop_seed);
        HIP_CHECK_LAST();
    }
}

}  // namespace

std::vector<torch::Tensor> flash_bwd(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                     torch::Tensor v, torch::Tensor out, torch::Tensor lse,
                                     c10::optional<torch::Tensor> pad_mask, bool causal,
                                     double dropout_p, int64_t seed) {
    TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
    dout = dout.contiguous(); q = q.contiguous(); k = k.contiguous(); v = v.contiguous();
    out = out.contiguous();
    int B = q.size(0), H = q.size(1), Nq = q.size(2), D = q.size(3);
    int Lk = k.size(2), Dv = v.size(3);

    auto delta = torch::empty({B, H, Nq}, q.options().dtype(torch::kFloat32));
    {
        long rows = (long)B * H * Nq;
        int waves_per_block = 4;
        long blocks = (rows + waves_per_block - 1) / waves_per_block;
        hipLaunchKernelGGL(delta_kernel, dim3(blocks), dim3(64 * waves_per_block), 0,
                           at::cuda::getCurrentCUDAStream(),
                           reinterpret_cast<const unsigned short*>(dout.data_ptr()),
                           reinterpret_cast<const unsigned short*>(out.data_ptr()),
                           delta.data_ptr<float>(), rows, Dv);
        HIP_CHECK_LAST();
    }

    auto dq = torch::empty_like(q);
    auto dk = torch::empty_like(k);
    auto dv = torch::empty_like(v);

    c10::optional<torch::Tensor> pm;
    if (pad_mask.has_value() && pad_mask->defined()) pm = pad_mask->contiguous();

    float dp = (float)dropout_p;
    unsigned long long sd = (unsigned long long)seed;
    if (D <= 32 && Dv <= 160)       launch_flash_bwd<32, 160>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else if (D <= 64 && Dv <= 64)   launch_flash_bwd<64, 64>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else if (D <= 128 && Dv <= 128) launch_flash_bwd<128, 128>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else if (D <= 160 && Dv <= 160) launch_flash_bwd<160, 160>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else                            launch_flash_bwd<352, 352>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);

    return {dq, dk, dv};
}
