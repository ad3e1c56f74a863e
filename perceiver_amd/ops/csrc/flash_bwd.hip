// Flash attention backward for CDNA4 (gfx950): recompute-based, two kernels
// (atomic-free): dQ parallel over Q blocks, dK/dV parallel over KV blocks.
//
//   delta_i = sum_c dO[i][c] * O[i][c]
//   P       = exp(QK^T - lse_i)            (masks + dropout reapplied)
//   dS      = P * (dO V^T * mask/(1-p) - delta_i)
//   dQ      = dS K          dK = dS^T Q          dV = (P*mask/(1-p))^T dO
//
// v4: QH 16-row A-fragment groups per wave (one B-fragment LDS read feeds QH
// MFMAs); transposed operands (K^T / Q^T / dO^T) live as 16-column subtiles
// written row-major and consumed through ds_read_b64_tr_b16 hardware transpose
// reads; per-template TILE sizes balance LDS occupancy vs the register budget.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <cfloat>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) short short4x;

namespace {

constexpr int NWAVES = 4;

// ---------------------------------------------------------------- delta kernel
__global__ void delta_kernel(const unsigned short* __restrict__ dout,
                             const unsigned short* __restrict__ out,
                             float* __restrict__ delta,
                             long dsb, long dsh, long dsn,  // dout strides
                             long osb, long osh, long osn,  // out strides
                             int H, int Nq, long rows, int dv) {
    long row = (long)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    int lane = threadIdx.x % 64;
    if (row >= rows) return;
    long bb = row / ((long)H * Nq);
    long rem = row % ((long)H * Nq);
    const unsigned short* a = dout + bb * dsb + (rem / Nq) * dsh + (rem % Nq) * dsn;
    const unsigned short* b = out + bb * osb + (rem / Nq) * osh + (rem % Nq) * osn;
    float acc = 0.f;
    for (int c = lane * 2; c + 1 < dv; c += 128) {
        acc += bf2f(a[c]) * bf2f(b[c]) + bf2f(a[c + 1]) * bf2f(b[c + 1]);
    }
    if (dv % 2 == 1 && lane == 0) acc += bf2f(a[dv - 1]) * bf2f(b[dv - 1]);
#pragma unroll
    for (int m = 1; m < 64; m <<= 1) acc += __shfl_xor(acc, m, 64);
    if (lane == 0) delta[row] = acc;
}

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// ---- 16-column-subtiled image + ds_read_b64_tr_b16 (see flash_fwd.hip) ----
// Replaces the separate transposed staging pass: the image is written ROW-major
// (one 16-B write per 8-channel granule) and consumed column-major through the
// hardware transpose read, so the 8-row packed-write pass (and its bank-conflict
// lockstep) disappears along with the duplicate global re-read.
template <int ROWS_TILE>
DEVINL void stage_sub16(const unsigned short* __restrict__ src, long src_stride,
                        int rows_valid, int d, int d_pad, char* lds, int tid) {
    constexpr int SUBE = ROWS_TILE * 16 + 8;
    const int gpr = d_pad / 8;
    const int total = ROWS_TILE * gpr;
    for (int g = tid; g < total; g += 256) {
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
        *reinterpret_cast<short8v*>(
            lds + ((c0 / 16) * SUBE + row * 16 + (c0 % 16)) * 2) = val;
    }
}

// B-fragment rows row0..row0+31, cols 16*sub..: per-lane 8-B loads of one 4x16
// row-major tile per 16-lane group; the instruction's fixed cross-lane exchange
// delivers column lo16.
template <int ROWS_TILE>
DEVINL bf16x8 read_bfrag_tr16(const char* lds, int sub, int row0, int hi4, int lo16) {
    constexpr int SUBE = ROWS_TILE * 16 + 8;
    const __bf16* base = reinterpret_cast<const __bf16*>(lds) +
                         sub * SUBE + (row0 + hi4 * 8) * 16 + lo16 * 4;
    auto pp = (__attribute__((address_space(3))) bf16x4*)base;
    bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(pp);
    bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(pp + 16);  // +4 rows
    bf16x8 out;
#pragma unroll
    for (int e = 0; e < 4; ++e) { out[e] = lo[e]; out[e + 4] = hi[e]; }
    return out;
}

// stage (rows_tile x d) tile into BOTH the row-major image (row stride
// ldst_bytes) and the 16-col subtiled image: one global read per granule,
// two LDS writes (replaces two separate staging passes = half the global
// load instructions on the q/do/k operands)
template <int ROWS_TILE>
DEVINL void stage_rm_sub16(const unsigned short* __restrict__ src, long src_stride,
                           int rows_valid, int d, int d_pad,
                           char* lds_rm, int ldst_bytes, char* lds16, int tid) {
    constexpr int SUBE = ROWS_TILE * 16 + 8;
    const int gpr = d_pad / 8;
    const int total = ROWS_TILE * gpr;
    for (int g = tid; g < total; g += 256) {
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
        *reinterpret_cast<short8v*>(lds_rm + row * ldst_bytes + c0 * 2) = val;
        *reinterpret_cast<short8v*>(
            lds16 + ((c0 / 16) * SUBE + row * 16 + (c0 % 16)) * 2) = val;
    }
}

// stage (rows_tile x d) tile row-major into LDS (row stride ldst_bytes), zero-pad
template <int ROWS_TILE>
DEVINL void stage_rm(const unsigned short* __restrict__ src, long src_stride,
                     int rows_valid, int d, int d_pad,
                     char* lds, int ldst_bytes, int tid) {
    const int gpr = d_pad / 8;
    const int total = ROWS_TILE * gpr;
    for (int g = tid; g < total; g += 256) {
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

// ---------------------------------------------------------------- dQ kernel
// grid.x over Q blocks (QH*16 rows per wave), grid.y = B*H. Stages per TILE-key
// tile: K row-major (for S), K^T (for dQ = dS K), V row-major (for dP = dO V^T).
template <int DMAX, int DVMAX, int TILE, int QH>
__launch_bounds__(256, DMAX <= 160 ? 2 : 1)
__global__ void flash_dq_kernel(
    const unsigned short* __restrict__ qp, const unsigned short* __restrict__ kp,
    const unsigned short* __restrict__ vp, const unsigned short* __restrict__ dop,
    const float* __restrict__ lsep, const float* __restrict__ deltap,
    const bool* __restrict__ pad,
    unsigned short* __restrict__ dqp,
    float* __restrict__ dq_part,  // (S,B,H,Nq,D) fp32 when gridDim.z > 1
    long kv_chunk,
    long qsb, long qsh, long qsn, long ksb, long ksh, long ksn,
    long vsb, long vsh, long vsn, long dsb, long dsh, long dsn,
    long oqsb, long oqsh, long oqsn,  // dq OUT strides (merged-heads memory)
    int B, int H, int Nq, int Lk, int D, int Dv, int causal,
    float drop_p, unsigned long long drop_seed) {
    constexpr int TBLKS = TILE / 16;
    constexpr int QROWS = 16 * QH;
    constexpr int QBLK = QROWS * NWAVES;
    const int d_pad = (D + 31) & ~31;
    const int dv_pad = (Dv + 31) & ~31;
    const int d_blocks = d_pad / 32;
    const int dv_blocks32 = dv_pad / 32;

    const int tid = threadIdx.x, wave = tid / 64, lane = tid % 64;
    const int lo16 = lane & 15, hi4 = lane >> 4;
    const int bh = blockIdx.y, b = bh / H, hh = bh % H;
    const int q0 = blockIdx.x * QBLK + wave * QROWS;

    const unsigned short* qbase = qp + (long)b * qsb + (long)hh * qsh;
    const unsigned short* kbase = kp + (long)b * ksb + (long)hh * ksh;
    const unsigned short* vbase = vp + (long)b * vsb + (long)hh * vsh;
    const unsigned short* dobase = dop + (long)b * dsb + (long)hh * dsh;
    const float* lse_row = lsep + (long)bh * Nq;
    const float* delta_row = deltap + (long)bh * Nq;
    const bool* padrow = pad ? pad + (long)b * Lk : nullptr;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int k_stride = d_pad * 2 + 16;
    const int kt_stride = TILE * 2 + 16;   // (p-buffer row stride)
    const int v_stride = dv_pad * 2 + 16;
    char* k_lds = smem;                                  // TILE * k_stride
    char* kt16_lds = k_lds + TILE * k_stride;            // (DMAX/16) * (TILE*16+8) elems
    char* v_lds = kt16_lds + (DMAX / 16) * (TILE * 16 + 8) * 2;   // TILE * v_stride
    // per-wave transposed dS image ([key][16 q-rows] subtile per h): packed
    // ushort2 writes + ds_read_b64_tr_b16 A-fragment reads (the row-major
    // image needed 32 scalar b16 stores per tile per wave)
    constexpr int SUBE_T = TILE * 16 + 8;
    char* p_lds = v_lds + TILE * v_stride;               // NWAVES * QH * SUBE_T * 2
    char* p_mine = p_lds + wave * QH * SUBE_T * 2;

    short8v q_frag[QH][DMAX / 32];
    short8v do_frag[QH][DVMAX / 32];
    float4v lse_r[QH], delta_r[QH];
#pragma unroll
    for (int h = 0; h < QH; ++h) {
        int qi = q0 + h * 16 + lo16;
        bool valid = qi < Nq;
        int qc = valid ? qi : Nq - 1;
        const unsigned short* qrow = qbase + (long)qc * qsn;
        const unsigned short* dorow = dobase + (long)qc * dsn;
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
            q_frag[h][kb] = val;
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
            do_frag[h][kb] = val;
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int qi2 = q0 + h * 16 + hi4 * 4 + r;
            lse_r[h][r] = (qi2 < Nq) ? lse_row[qi2] : 0.f;
            delta_r[h][r] = (qi2 < Nq) ? delta_row[qi2] : 0.f;  // float4v lanes
        }
    }

    float4v dq_acc[QH][DMAX / 16];
#pragma unroll
    for (int h = 0; h < QH; ++h)
#pragma unroll
        for (int cb = 0; cb < DMAX / 16; ++cb) dq_acc[h][cb] = float4v{0.f, 0.f, 0.f, 0.f};

    int kv_end = Lk;
    if (causal) kv_end = min(Lk, Lk - Nq + blockIdx.x * QBLK + QBLK);
    int kv_begin = 0;
    if (gridDim.z > 1) {  // KV-split; fp32 partials summed by the caller
        kv_begin = (int)((long)blockIdx.z * kv_chunk);
        kv_end = min((long)kv_end, (long)(blockIdx.z + 1) * kv_chunk);
    }
    const unsigned int drop_thresh = (unsigned int)(drop_p * 65536.0);

    // T14 split staging on exact template matches: K/V loads for tile t+1 fly
    // during tile t's compute instead of serializing between the barriers
    // dkv stays on the slow stage for the wide pad tiers: fast staging at
    // 288/352 spilled 37-298 VGPRs (this kernel's accumulators already fill
    // the file at TILE=32)
    constexpr bool kFast = (DMAX % 32 == 0) && (DVMAX % 32 == 0) && (DMAX <= 64);
    constexpr int NG_K = kFast ? (TILE * DMAX + 2047) / 2048 : 1;
    constexpr int NG_V = kFast ? (TILE * DVMAX + 2047) / 2048 : 1;
    short8v st_k[NG_K], st_v[NG_V];
    const bool fast = kFast && D == DMAX && Dv == DVMAX;
    const int n_tiles = (kv_end > kv_begin) ? (kv_end - kv_begin + TILE - 1) / TILE : 0;
    const int kv_last = kv_begin + (n_tiles > 0 ? (n_tiles - 1) * TILE : 0);
    if (fast && n_tiles > 0) {
        issue_tile<TILE, DMAX>(st_k, kbase + (long)kv_begin * ksn, ksn, Lk - kv_begin, tid);
        issue_tile<TILE, DVMAX>(st_v, vbase + (long)kv_begin * vsn, vsn, Lk - kv_begin, tid);
    }

    for (int kv0 = kv_begin; kv0 < kv_end; kv0 += TILE) {
        int rows_valid = min(TILE, Lk - kv0);
        __syncthreads();
        if (fast) {
            write_rm_sub16_c<TILE, DMAX>(st_k, k_lds, k_stride, kt16_lds, rows_valid, tid);
            write_rm<TILE, DVMAX>(st_v, v_lds, v_stride, rows_valid, tid);
            int kv_n = min(kv0 + TILE, kv_last);
            issue_tile<TILE, DMAX>(st_k, kbase + (long)kv_n * ksn, ksn, Lk - kv_n, tid);
            issue_tile<TILE, DVMAX>(st_v, vbase + (long)kv_n * vsn, vsn, Lk - kv_n, tid);
        } else {
            stage_rm_sub16<TILE>(kbase + (long)kv0 * ksn, ksn, rows_valid, D, d_pad,
                                 k_lds, k_stride, kt16_lds, tid);
            stage_rm<TILE>(vbase + (long)kv0 * vsn, vsn, rows_valid, Dv, dv_pad, v_lds, v_stride, tid);
        }
        __syncthreads();
        // wave-uniform mask hoist (see flash_fwd.hip): interior tiles of the
        // no-pad case skip the per-element mask chain entirely
        const bool tile_masked =
            (kv0 + TILE > Lk) || (padrow != nullptr) ||
            (causal && kv0 + TILE - 1 > Lk - Nq + q0);

        // t-outer: per 16-key block compute S and dP with short-lived accumulators,
        // convert to dS and spill to the per-wave LDS buffer immediately
#pragma unroll
        for (int t = 0; t < TBLKS; ++t) {
            float4v s_acc[QH], dp_acc[QH];
#pragma unroll
            for (int h = 0; h < QH; ++h) {
                s_acc[h] = float4v{0.f, 0.f, 0.f, 0.f};
                dp_acc[h] = float4v{0.f, 0.f, 0.f, 0.f};
            }
#pragma unroll
            for (int kb = 0; kb < DMAX / 32; ++kb) {
                if (kb < d_blocks) {
                    const char* src = k_lds + (t * 16 + lo16) * k_stride + (kb * 32 + hi4 * 8) * 2;
                    bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
#pragma unroll
                    for (int h = 0; h < QH; ++h)
                        s_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            (bf16x8)q_frag[h][kb], bfrag, s_acc[h], 0, 0, 0);
                }
            }
#pragma unroll
            for (int kb = 0; kb < DVMAX / 32; ++kb) {
                if (kb < dv_blocks32) {
                    const char* src = v_lds + (t * 16 + lo16) * v_stride + (kb * 32 + hi4 * 8) * 2;
                    bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
#pragma unroll
                    for (int h = 0; h < QH; ++h)
                        dp_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            (bf16x8)do_frag[h][kb], bfrag, dp_acc[h], 0, 0, 0);
                }
            }
#pragma unroll
            for (int h = 0; h < QH; ++h) {
                float ds_pack[4];
                // packed fp32 dS math (v_pk halves the VALU); the dropout
                // scale folds into the per-element multiplier m4
                float4v e4 = s_acc[h] - lse_r[h];
                float4v p4;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int qi = q0 + h * 16 + hi4 * 4 + r;
                    int j = kv0 + t * 16 + lo16;
                    bool masked = tile_masked &&
                                  (j >= Lk || (padrow && j < Lk && padrow[j]) ||
                                   (causal && j > Lk - Nq + qi));
                    p4[r] = masked ? 0.f : __expf(e4[r]);
                }
                float4v m4 = {1.f, 1.f, 1.f, 1.f};
                if (drop_p > 0.f) {
                    // explicit qi-pair hashes (see common.h drop16 mapping)
                    int j = kv0 + t * 16 + lo16;
                    int qb2 = (q0 + h * 16 + hi4 * 4) >> 1;
                    unsigned int hh[2] = {rng_hash(drop_seed, bh, qb2, j),
                                          rng_hash(drop_seed, bh, qb2 + 1, j)};
                    const float inv_keep = 1.0f / (1.0f - drop_p);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        unsigned int d = (r & 1) ? (hh[r >> 1] >> 16) : (hh[r >> 1] & 0xffffu);
                        m4[r] = (d >= drop_thresh) ? inv_keep : 0.f;
                    }
                }
                float4v ds4 = p4 * (dp_acc[h] * m4 - delta_r[h]);
#pragma unroll
                for (int r = 0; r < 4; ++r) ds_pack[r] = ds4[r];
                unsigned short* dst = reinterpret_cast<unsigned short*>(
                    p_mine + (h * SUBE_T + (t * 16 + lo16) * 16 + hi4 * 4) * 2);
                *reinterpret_cast<uint2v*>(dst) =
                    f2bf4(ds_pack[0], ds_pack[1], ds_pack[2], ds_pack[3]);
            }
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        bf16x8 ds_frag[QH][TBLKS / 2];
#pragma unroll
        for (int h = 0; h < QH; ++h)
#pragma unroll
            for (int t32 = 0; t32 < TBLKS / 2; ++t32)
                ds_frag[h][t32] = read_bfrag_tr16<TILE>(
                    p_mine + h * SUBE_T * 2, 0, t32 * 32, hi4, lo16);

        // dQ += dS K : B[k=key][j=ch] via transpose reads of the subtiled K image
#pragma unroll
        for (int cb = 0; cb < DMAX / 16; ++cb) {
            if (cb * 16 < d_pad) {
#pragma unroll
                for (int t32 = 0; t32 < TBLKS / 2; ++t32) {
                    bf16x8 bfrag = read_bfrag_tr16<TILE>(kt16_lds, cb, t32 * 32, hi4, lo16);
#pragma unroll
                    for (int h = 0; h < QH; ++h)
                        dq_acc[h][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            ds_frag[h][t32], bfrag, dq_acc[h][cb], 0, 0, 0);
                }
            }
        }
    }

    // store dQ (C layout rows h*16 + hi4*4+r, col lo16+16cb); fp32 partial per
    // split when KV-split is active (caller sums)
#pragma unroll
    for (int h = 0; h < QH; ++h)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int qi = q0 + h * 16 + hi4 * 4 + r;
            if (qi >= Nq) continue;
            if (gridDim.z > 1) {
                float* dqrow = dq_part + (((long)blockIdx.z * B * H + bh) * Nq + qi) * D;
#pragma unroll
                for (int cb = 0; cb < DMAX / 16; ++cb) {
                    int c = cb * 16 + lo16;
                    if (c < D) dqrow[c] = dq_acc[h][cb][r];
                }
            } else {
                unsigned short* dqrow = dqp + (long)b * oqsb + (long)hh * oqsh + (long)qi * oqsn;
#pragma unroll
                for (int cb = 0; cb < DMAX / 16; ++cb) {
                    int c = cb * 16 + lo16;
                    if (c < D) dqrow[c] = f2bf(dq_acc[h][cb][r]);
                }
            }
        }
}

// ---------------------------------------------------------------- dK/dV kernel
// grid.x over KV blocks (QH*16 keys per wave), grid.y = B*H.
// Loops over TILE-row Q tiles; stages Q row-major + Q^T, dO row-major + dO^T.
template <int DMAX, int DVMAX, int TILE, int QH>
__launch_bounds__(256, DMAX <= 128 ? 2 : 1)
__global__ void flash_dkv_kernel(
    const unsigned short* __restrict__ qp, const unsigned short* __restrict__ kp,
    const unsigned short* __restrict__ vp, const unsigned short* __restrict__ dop,
    const float* __restrict__ lsep, const float* __restrict__ deltap,
    const bool* __restrict__ pad,
    unsigned short* __restrict__ dkp, unsigned short* __restrict__ dvp,
    float* __restrict__ dk_part, float* __restrict__ dv_part,  // (S,B*H,Lk,D[v]) when gridDim.z>1
    long q_chunk,
    long qsb, long qsh, long qsn, long ksb, long ksh, long ksn,
    long vsb, long vsh, long vsn, long dsb, long dsh, long dsn,
    long oksb, long oksh, long oksn, long ovsb, long ovsh, long ovsn,  // dk/dv OUT strides
    int B, int H, int Nq, int Lk, int D, int Dv, int causal,
    float drop_p, unsigned long long drop_seed) {
    constexpr int TBLKS = TILE / 16;
    constexpr int KROWS = 16 * QH;
    constexpr int KBLK = KROWS * NWAVES;
    const int d_pad = (D + 31) & ~31;
    const int dv_pad = (Dv + 31) & ~31;
    const int d_blocks = d_pad / 32;
    const int dv_blocks32 = dv_pad / 32;

    const int tid = threadIdx.x, wave = tid / 64, lane = tid % 64;
    const int lo16 = lane & 15, hi4 = lane >> 4;
    const int bh = blockIdx.y, b = bh / H, hh = bh % H;
    const int k0 = blockIdx.x * KBLK + wave * KROWS;  // this wave's first key row

    const unsigned short* qbase = qp + (long)b * qsb + (long)hh * qsh;
    const unsigned short* kbase = kp + (long)b * ksb + (long)hh * ksh;
    const unsigned short* vbase = vp + (long)b * vsb + (long)hh * vsh;
    const unsigned short* dobase = dop + (long)b * dsb + (long)hh * dsh;
    const float* lse_row = lsep + (long)bh * Nq;
    const float* delta_row = deltap + (long)bh * Nq;
    const bool* padrow = pad ? pad + (long)b * Lk : nullptr;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int q_stride = d_pad * 2 + 16;      // Q row-major: TILE rows
    const int qt_stride = TILE * 2 + 16;      // (p-buffer row stride)
    const int do_stride = dv_pad * 2 + 16;    // dO row-major: TILE rows
    char* q_lds = smem;
    char* q16_lds = q_lds + TILE * q_stride;              // (DMAX/16)*(TILE*16+8) elems
    char* do_lds = q16_lds + (DMAX / 16) * (TILE * 16 + 8) * 2;
    char* do16_lds = do_lds + TILE * do_stride;           // (DVMAX/16)*(TILE*16+8) elems
    // per-wave transposed P/dS image ([q][16 key-rows] subtile per h) —
    // packed ushort2 writes + tr16 A-fragment reads, see the dq kernel
    constexpr int SUBE_T = TILE * 16 + 8;
    // two per-wave image halves: P and dS written back-to-back, ONE lgkm
    // wait covers both reads (the second roundtrip stall is gone)
    char* p_lds = do16_lds + (DVMAX / 16) * (TILE * 16 + 8) * 2;  // NWAVES*2*QH*SUBE_T*2
    char* p_mine = p_lds + wave * 2 * QH * SUBE_T * 2;
    char* ds_mine = p_mine + QH * SUBE_T * 2;

    short8v k_frag[QH][DMAX / 32];
    short8v v_frag[QH][DVMAX / 32];
    bool key_pad[QH][4];
#pragma unroll
    for (int h = 0; h < QH; ++h) {
        int ki = k0 + h * 16 + lo16;
        bool valid = ki < Lk;
        int kc = valid ? ki : Lk - 1;
        const unsigned short* krow = kbase + (long)kc * ksn;
        const unsigned short* vrow = vbase + (long)kc * vsn;
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
            k_frag[h][kb] = val;
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
            v_frag[h][kb] = val;
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int ki2 = k0 + h * 16 + hi4 * 4 + r;
            key_pad[h][r] = (ki2 >= Lk) || (padrow && padrow[min(ki2, Lk - 1)]);
        }
    }
    bool any_keypad = false;
#pragma unroll
    for (int h = 0; h < QH; ++h)
#pragma unroll
        for (int r = 0; r < 4; ++r) any_keypad |= key_pad[h][r];
    any_keypad = __any(any_keypad);

    float4v dk_acc[QH][DMAX / 16];
    float4v dv_acc[QH][DVMAX / 16];
#pragma unroll
    for (int h = 0; h < QH; ++h) {
#pragma unroll
        for (int cb = 0; cb < DMAX / 16; ++cb) dk_acc[h][cb] = float4v{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int cb = 0; cb < DVMAX / 16; ++cb) dv_acc[h][cb] = float4v{0.f, 0.f, 0.f, 0.f};
    }

    int q_start = 0;
    if (causal) {
        int j_lo = blockIdx.x * KBLK;
        q_start = max(0, j_lo - (Lk - Nq));
        q_start = (q_start / TILE) * TILE;
    }
    // Q-split over gridDim.z for grids that underfill the chip (e.g. the
    // flow/img DECODER cross-attention backward: 50k-182k queries against a
    // few hundred latent keys = a handful of key blocks); fp32 partials
    // summed by the launcher
    int q_end = Nq;
    if (gridDim.z > 1) {
        q_start = max(q_start, (int)((long)blockIdx.z * q_chunk));
        q_end = (int)min((long)Nq, (long)(blockIdx.z + 1) * q_chunk);
    }
    const unsigned int drop_thresh = (unsigned int)(drop_p * 65536.0);

    // T14 split staging (see flash_dq_kernel): Q/dO tile t+1 loads fly under
    // tile t's MFMA work on exact template matches
    // 352 measured SLOWER with fast staging (96 staged VGPRs at occupancy 1
    // starve the S/dP pipeline); 288 gains — img encoder-CA backward
    constexpr bool kFast = (DMAX % 32 == 0) && (DVMAX % 32 == 0) &&
                           (DMAX <= 128 || DMAX == 288);
    constexpr int NG_Q = kFast ? (TILE * DMAX + 2047) / 2048 : 1;
    constexpr int NG_DO = kFast ? (TILE * DVMAX + 2047) / 2048 : 1;
    short8v st_q[NG_Q], st_do[NG_DO];
    const bool fast = kFast && D == DMAX && Dv == DVMAX;
    const int nq_tiles = (q_end > q_start) ? (q_end - q_start + TILE - 1) / TILE : 0;
    const int qt_last = q_start + (nq_tiles > 0 ? (nq_tiles - 1) * TILE : 0);
    if (fast && nq_tiles > 0) {
        issue_tile<TILE, DMAX>(st_q, qbase + (long)q_start * qsn, qsn, Nq - q_start, tid);
        issue_tile<TILE, DVMAX>(st_do, dobase + (long)q_start * dsn, dsn, Nq - q_start, tid);
    }

    for (int qt0 = q_start; qt0 < q_end; qt0 += TILE) {
        int rows_valid = min(TILE, Nq - qt0);
        __syncthreads();
        if (fast) {
            write_rm_sub16_c<TILE, DMAX>(st_q, q_lds, q_stride, q16_lds, rows_valid, tid);
            write_rm_sub16_c<TILE, DVMAX>(st_do, do_lds, do_stride, do16_lds, rows_valid, tid);
            int qt_n = min(qt0 + TILE, qt_last);
            issue_tile<TILE, DMAX>(st_q, qbase + (long)qt_n * qsn, qsn, Nq - qt_n, tid);
            issue_tile<TILE, DVMAX>(st_do, dobase + (long)qt_n * dsn, dsn, Nq - qt_n, tid);
        } else {
            stage_rm_sub16<TILE>(qbase + (long)qt0 * qsn, qsn, rows_valid, D, d_pad,
                                 q_lds, q_stride, q16_lds, tid);
            stage_rm_sub16<TILE>(dobase + (long)qt0 * dsn, dsn, rows_valid, Dv, dv_pad,
                                 do_lds, do_stride, do16_lds, tid);
        }
        __syncthreads();
        // wave-uniform mask hoist: interior q-tiles with no padded/overhang
        // keys skip the per-element mask chain (kernels are VALU/wait-bound)
        const bool tile_masked =
            any_keypad || (qt0 + TILE > Nq) ||
            (causal && k0 + KROWS - 1 > Lk - Nq + qt0);

        // t-outer: per 16-q-row block compute S^T and dP^T with short-lived
        // accumulators; P^T goes to LDS now, dS^T is kept in a small register array
#pragma unroll
        for (int t = 0; t < TBLKS; ++t) {
            float4v st_acc[QH], dpt_acc[QH];
#pragma unroll
            for (int h = 0; h < QH; ++h) {
                st_acc[h] = float4v{0.f, 0.f, 0.f, 0.f};
                dpt_acc[h] = float4v{0.f, 0.f, 0.f, 0.f};
            }
#pragma unroll
            for (int kb = 0; kb < DMAX / 32; ++kb) {
                if (kb < d_blocks) {
                    const char* src = q_lds + (t * 16 + lo16) * q_stride + (kb * 32 + hi4 * 8) * 2;
                    bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
#pragma unroll
                    for (int h = 0; h < QH; ++h)
                        st_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            (bf16x8)k_frag[h][kb], bfrag, st_acc[h], 0, 0, 0);
                }
            }
#pragma unroll
            for (int kb = 0; kb < DVMAX / 32; ++kb) {
                if (kb < dv_blocks32) {
                    const char* src = do_lds + (t * 16 + lo16) * do_stride + (kb * 32 + hi4 * 8) * 2;
                    bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
#pragma unroll
                    for (int h = 0; h < QH; ++h)
                        dpt_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            (bf16x8)v_frag[h][kb], bfrag, dpt_acc[h], 0, 0, 0);
                }
            }
#pragma unroll
            for (int h = 0; h < QH; ++h) {
                // qi (and so lse/delta) is fixed across the 4 key rows, so the
                // dS/P math runs on packed fp32 (v_pk_mul/add halve the VALU);
                // the dropout scale folds into a per-element multiplier m4
                const int qi = qt0 + t * 16 + lo16;
                const float lse_i = (qi < Nq) ? lse_row[qi] : 0.f;
                const float delta_i = (qi < Nq) ? delta_row[qi] : 0.f;
                float4v e4 = st_acc[h] - lse_i;
                float4v p4;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int ki = k0 + h * 16 + hi4 * 4 + r;
                    bool masked = tile_masked &&
                                  (key_pad[h][r] || qi >= Nq || (causal && ki > Lk - Nq + qi));
                    p4[r] = masked ? 0.f : __expf(e4[r]);
                }
                float4v m4 = {1.f, 1.f, 1.f, 1.f};
                if (drop_p > 0.f) {
                    const float inv_keep = 1.0f / (1.0f - drop_p);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int ki = k0 + h * 16 + hi4 * 4 + r;
                        m4[r] = (drop16(drop_seed, bh, qi, ki) >= drop_thresh) ? inv_keep : 0.f;
                    }
                }
                float4v ds4 = p4 * (dpt_acc[h] * m4 - delta_i);
                float4v pe4 = p4 * m4;
                const long ioff = (h * SUBE_T + (t * 16 + lo16) * 16 + hi4 * 4) * 2;
                *reinterpret_cast<uint2v*>(p_mine + ioff) =
                    f2bf4(pe4[0], pe4[1], pe4[2], pe4[3]);
                *reinterpret_cast<uint2v*>(ds_mine + ioff) =
                    f2bf4(ds4[0], ds4[1], ds4[2], ds4[3]);
            }
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        bf16x8 pt_frag[QH][TBLKS / 2], dst_frag[QH][TBLKS / 2];
#pragma unroll
        for (int h = 0; h < QH; ++h)
#pragma unroll
            for (int t32 = 0; t32 < TBLKS / 2; ++t32) {
                pt_frag[h][t32] = read_bfrag_tr16<TILE>(
                    p_mine + h * SUBE_T * 2, 0, t32 * 32, hi4, lo16);
                dst_frag[h][t32] = read_bfrag_tr16<TILE>(
                    ds_mine + h * SUBE_T * 2, 0, t32 * 32, hi4, lo16);
            }

        // dV += P^T dO : B[k=qrow][j=ch] via transpose reads of the subtiled image
#pragma unroll
        for (int cb = 0; cb < DVMAX / 16; ++cb) {
            if (cb * 16 < dv_pad) {
#pragma unroll
                for (int t32 = 0; t32 < TBLKS / 2; ++t32) {
                    bf16x8 bfrag = read_bfrag_tr16<TILE>(do16_lds, cb, t32 * 32, hi4, lo16);
#pragma unroll
                    for (int h = 0; h < QH; ++h)
                        dv_acc[h][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            pt_frag[h][t32], bfrag, dv_acc[h][cb], 0, 0, 0);
                }
            }
        }
        // dK += dS^T Q : B[k=qrow][j=ch] via transpose reads of the subtiled image
#pragma unroll
        for (int cb = 0; cb < DMAX / 16; ++cb) {
            if (cb * 16 < d_pad) {
#pragma unroll
                for (int t32 = 0; t32 < TBLKS / 2; ++t32) {
                    bf16x8 bfrag = read_bfrag_tr16<TILE>(q16_lds, cb, t32 * 32, hi4, lo16);
#pragma unroll
                    for (int h = 0; h < QH; ++h)
                        dk_acc[h][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            dst_frag[h][t32], bfrag, dk_acc[h][cb], 0, 0, 0);
                }
            }
        }
    }

    // store dK/dV (C layout: key row h*16 + hi4*4+r, col lo16+16cb);
    // fp32 partial slabs when Q-split (summed by the launcher)
#pragma unroll
    for (int h = 0; h < QH; ++h)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int ki = k0 + h * 16 + hi4 * 4 + r;
            if (ki >= Lk) continue;
            if (gridDim.z > 1) {
                long base = ((long)blockIdx.z * B * H + bh) * Lk + ki;
                float* dkrow = dk_part + base * D;
                float* dvrow = dv_part + base * Dv;
#pragma unroll
                for (int cb = 0; cb < DMAX / 16; ++cb) {
                    int c = cb * 16 + lo16;
                    if (c < D) dkrow[c] = dk_acc[h][cb][r];
                }
#pragma unroll
                for (int cb = 0; cb < DVMAX / 16; ++cb) {
                    int c = cb * 16 + lo16;
                    if (c < Dv) dvrow[c] = dv_acc[h][cb][r];
                }
            } else {
                unsigned short* dkrow = dkp + (long)b * oksb + (long)hh * oksh + (long)ki * oksn;
                unsigned short* dvrow = dvp + (long)b * ovsb + (long)hh * ovsh + (long)ki * ovsn;
#pragma unroll
                for (int cb = 0; cb < DMAX / 16; ++cb) {
                    int c = cb * 16 + lo16;
                    if (c < D) dkrow[c] = f2bf(dk_acc[h][cb][r]);
                }
#pragma unroll
                for (int cb = 0; cb < DVMAX / 16; ++cb) {
                    int c = cb * 16 + lo16;
                    if (c < Dv) dvrow[c] = f2bf(dv_acc[h][cb][r]);
                }
            }
        }
}

template <int DMAX, int DVMAX, int DQ_TILE, int DQ_QH, int DKV_TILE, int DKV_QH>
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

    {   // dQ (with KV-split when the grid would underfill the chip)
        const int k_stride = d_pad * 2 + 16, kt_stride = DQ_TILE * 2 + 16, v_stride = dv_pad * 2 + 16;
        const int qblk = 16 * DQ_QH * NWAVES;
        size_t smem = (size_t)DQ_TILE * k_stride + (size_t)(DMAX / 16) * (DQ_TILE * 16 + 8) * 2 +
                      (size_t)DQ_TILE * v_stride + (size_t)NWAVES * DQ_QH * (DQ_TILE * 16 + 8) * 2;
        int gx = (Nq + qblk - 1) / qblk, gy = B * H;
        int nsplit = 1;
        long kv_chunk = Lk;
        if ((long)gx * gy < 512 && Lk > 4 * DQ_TILE) {
            int want = (int)(512 / ((long)gx * gy)) + 1;
            int max_split = (Lk + 4 * DQ_TILE - 1) / (4 * DQ_TILE);
            nsplit = std::min({want, max_split, 32});
            long tiles = (Lk + DQ_TILE - 1) / DQ_TILE;
            long tiles_per = (tiles + nsplit - 1) / nsplit;
            kv_chunk = tiles_per * DQ_TILE;
            nsplit = (int)((Lk + kv_chunk - 1) / kv_chunk);
        }
        torch::Tensor dq_part;
        float* dq_part_p = nullptr;
        if (nsplit > 1) {
            dq_part = torch::empty({(long)nsplit, (long)B * H * Nq, (long)D},
                                   q.options().dtype(torch::kFloat32));
            dq_part_p = dq_part.data_ptr<float>();
        }
        dim3 grid(gx, gy, nsplit);
        hipLaunchKernelGGL((flash_dq_kernel<DMAX, DVMAX, DQ_TILE, DQ_QH>), grid, dim3(256), smem, stream,
                           reinterpret_cast<const unsigned short*>(q.data_ptr()),
                           reinterpret_cast<const unsigned short*>(k.data_ptr()),
                           reinterpret_cast<const unsigned short*>(v.data_ptr()),
                           reinterpret_cast<const unsigned short*>(dout.data_ptr()),
                           lse.data_ptr<float>(), delta.data_ptr<float>(), padp,
                           reinterpret_cast<unsigned short*>(dq.data_ptr()),
                           dq_part_p, kv_chunk,
                           q.stride(0), q.stride(1), q.stride(2),
                           k.stride(0), k.stride(1), k.stride(2),
                           v.stride(0), v.stride(1), v.stride(2),
                           dout.stride(0), dout.stride(1), dout.stride(2),
                           dq.stride(0), dq.stride(1), dq.stride(2),
                           B, H, Nq, Lk, D, Dv, (int)causal, drop_p, drop_seed);
        HIP_CHECK_LAST();
        if (nsplit > 1) {
            dq.copy_(dq_part.sum(0).view_as(dq));
        }
    }
    {   // dK/dV (with Q-split when the key grid underfills the chip)
        const int q_stride = d_pad * 2 + 16, qt_stride = DKV_TILE * 2 + 16;
        const int do_stride = dv_pad * 2 + 16;
        const int kblk = 16 * DKV_QH * NWAVES;
        size_t smem = (size_t)DKV_TILE * q_stride + (size_t)(DMAX / 16) * (DKV_TILE * 16 + 8) * 2 +
                      (size_t)DKV_TILE * do_stride + (size_t)(DVMAX / 16) * (DKV_TILE * 16 + 8) * 2 +
                      (size_t)NWAVES * 2 * DKV_QH * (DKV_TILE * 16 + 8) * 2;
        int gx = (Lk + kblk - 1) / kblk, gy = B * H;
        int nsplit = 1;
        long q_chunk = Nq;
        if (!causal && (long)gx * gy < 512 && Nq > 4 * DKV_TILE) {
            int want = (int)(512 / ((long)gx * gy)) + 1;
            int max_split = (Nq + 4 * DKV_TILE - 1) / (4 * DKV_TILE);
            nsplit = std::min({want, max_split, 32});
            long tiles = (Nq + DKV_TILE - 1) / DKV_TILE;
            long tiles_per = (tiles + nsplit - 1) / nsplit;
            q_chunk = tiles_per * DKV_TILE;
            nsplit = (int)((Nq + q_chunk - 1) / q_chunk);
        }
        torch::Tensor dk_part, dv_part;
        float* dk_part_p = nullptr;
        float* dv_part_p = nullptr;
        if (nsplit > 1) {
            dk_part = torch::zeros({(long)nsplit, (long)B * H, (long)Lk, (long)D},
                                   q.options().dtype(torch::kFloat32));
            dv_part = torch::zeros({(long)nsplit, (long)B * H, (long)Lk, (long)Dv},
                                   q.options().dtype(torch::kFloat32));
            dk_part_p = dk_part.data_ptr<float>();
            dv_part_p = dv_part.data_ptr<float>();
        }
        dim3 grid(gx, gy, nsplit);
        if (smem > 65536) {
            static bool raised = [] {
                (void)hipFuncSetAttribute(
                    reinterpret_cast<const void*>(&flash_dkv_kernel<DMAX, DVMAX, DKV_TILE, DKV_QH>),
                    hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
                return true;
            }();
            (void)raised;
        }
        hipLaunchKernelGGL((flash_dkv_kernel<DMAX, DVMAX, DKV_TILE, DKV_QH>), grid, dim3(256), smem, stream,
                           reinterpret_cast<const unsigned short*>(q.data_ptr()),
                           reinterpret_cast<const unsigned short*>(k.data_ptr()),
                           reinterpret_cast<const unsigned short*>(v.data_ptr()),
                           reinterpret_cast<const unsigned short*>(dout.data_ptr()),
                           lse.data_ptr<float>(), delta.data_ptr<float>(), padp,
                           reinterpret_cast<unsigned short*>(dk.data_ptr()),
                           reinterpret_cast<unsigned short*>(dv.data_ptr()),
                           dk_part_p, dv_part_p, q_chunk,
                           q.stride(0), q.stride(1), q.stride(2),
                           k.stride(0), k.stride(1), k.stride(2),
                           v.stride(0), v.stride(1), v.stride(2),
                           dout.stride(0), dout.stride(1), dout.stride(2),
                           dk.stride(0), dk.stride(1), dk.stride(2),
                           dv.stride(0), dv.stride(1), dv.stride(2),
                           B, H, Nq, Lk, D, Dv, (int)causal, drop_p, drop_seed);
        HIP_CHECK_LAST();
        if (nsplit > 1) {
            dk.copy_(dk_part.sum(0).view_as(dk));
            dv.copy_(dv_part.sum(0).view_as(dv));
        }
    }
}

}  // namespace

std::vector<torch::Tensor> flash_bwd(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                     torch::Tensor v, torch::Tensor out, torch::Tensor lse,
                                     c10::optional<torch::Tensor> pad_mask, bool causal,
                                     double dropout_p, int64_t seed) {
    TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
    // last-dim contiguity suffices; the merged-heads (B,N,H,Dv) layouts the
    // forward now produces (and the matching grads) pass through zero-copy
    if (dout.stride(3) != 1) dout = dout.contiguous();
    if (out.stride(3) != 1) out = out.contiguous();
    if (q.stride(3) != 1) q = q.contiguous();
    if (k.stride(3) != 1) k = k.contiguous();
    if (v.stride(3) != 1) v = v.contiguous();
    int B = q.size(0), H = q.size(1), Nq = q.size(2), D = q.size(3);
    int Lk = k.size(2), Dv = v.size(3);

    // mirror the forward's pad-to-288 shim (see flash_fwd.hip): zero-padded
    // channels are exact — padded dout/out columns contribute 0 to delta, and
    // the padded dq/dk/dv columns are sliced off
    if (D > 160 && (D != 288 || Dv != 288) && D <= 288 && Dv <= 288) {
        namespace F = torch::nn::functional;
        auto res = flash_bwd(F::pad(dout, F::PadFuncOptions({0, 288 - Dv})),
                             F::pad(q, F::PadFuncOptions({0, 288 - D})),
                             F::pad(k, F::PadFuncOptions({0, 288 - D})),
                             F::pad(v, F::PadFuncOptions({0, 288 - Dv})),
                             F::pad(out, F::PadFuncOptions({0, 288 - Dv})),
                             lse, pad_mask, causal, dropout_p, seed);
        return {res[0].narrow(-1, 0, D), res[1].narrow(-1, 0, D), res[2].narrow(-1, 0, Dv)};
    }

    auto delta = torch::empty({B, H, Nq}, q.options().dtype(torch::kFloat32));
    {
        long rows = (long)B * H * Nq;
        int waves_per_block = 4;
        long blocks = (rows + waves_per_block - 1) / waves_per_block;
        hipLaunchKernelGGL(delta_kernel, dim3(blocks), dim3(64 * waves_per_block), 0,
                           at::cuda::getCurrentCUDAStream(),
                           reinterpret_cast<const unsigned short*>(dout.data_ptr()),
                           reinterpret_cast<const unsigned short*>(out.data_ptr()),
                           delta.data_ptr<float>(),
                           dout.stride(0), dout.stride(1), dout.stride(2),
                           out.stride(0), out.stride(1), out.stride(2),
                           H, Nq, rows, Dv);
        HIP_CHECK_LAST();
    }

    // grads live in merged-heads (B, N, H*D) memory and are returned as the
    // (B, H, N, D) permuted views: the module's head-merge transpose+reshape
    // on each grad becomes a free view instead of a 40 MB copy per tensor
    // (the forward's out uses the same trick)
    auto dq = torch::empty({q.size(0), q.size(2), q.size(1), q.size(3)}, q.options())
                  .permute({0, 2, 1, 3});
    auto dk = torch::empty({k.size(0), k.size(2), k.size(1), k.size(3)}, k.options())
                  .permute({0, 2, 1, 3});
    auto dv = torch::empty({v.size(0), v.size(2), v.size(1), v.size(3)}, v.options())
                  .permute({0, 2, 1, 3});

    c10::optional<torch::Tensor> pm;
    if (pad_mask.has_value() && pad_mask->defined()) pm = pad_mask->contiguous();

    float dp = (float)dropout_p;
    unsigned long long sd = (unsigned long long)seed;
    // <DMAX, DVMAX, DQ_TILE, DQ_QH, DKV_TILE, DKV_QH> — tiles/QH balance LDS
    // occupancy against the 256-VGPR budget per template
    if (D <= 32 && Dv <= 96)
        launch_flash_bwd<32, 96, 64, 2, 64, 1>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else if (D <= 32 && Dv <= 160)
        launch_flash_bwd<32, 160, 64, 2, 64, 1>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else if (D <= 64 && Dv <= 64)
        launch_flash_bwd<64, 64, 64, 2, 64, 1>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else if (D <= 128 && Dv <= 128)
        launch_flash_bwd<128, 128, 64, 2, 32, 1>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else if (D <= 160 && Dv <= 160)
        launch_flash_bwd<160, 160, 64, 1, 32, 1>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else if (D <= 288 && Dv <= 288)
        // the img/flow cross-attention class (D = 261, pad 288): dedicated
        // instantiation — the 352 template allocated for 352-wide register
        // arrays and spilled 19 B/lane on dkv
        launch_flash_bwd<288, 288, 32, 1, 32, 1>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);
    else
        launch_flash_bwd<352, 352, 32, 1, 32, 1>(dout, q, k, v, lse, delta, pm, causal, dp, sd, dq, dk, dv);

    return {dq, dk, dv};
}
