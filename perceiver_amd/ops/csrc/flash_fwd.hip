// Fused flash-style attention forward for CDNA4 (gfx950) — SURVEY.md §2.3 K1/K3/K4/K7.
//
// Handles every Perceiver attention regime with one templated kernel:
//   - encoder cross-attention  (short Q / long KV, e.g. 512 x 50k, odd D like 261)
//   - latent self-attention    (N x N, D 32..160)
//   - Perceiver-AR causal cross/self attention (right-aligned causal mask)
//   - decoder cross-attention  (long Q / short KV)
//
// Semantics match perceiver_amd.ops.attention.eager_attention: q arrives pre-scaled,
// pad_mask (B, Lk) bool True=pad masked with -FLT_MAX (so fully-masked rows degrade
// to uniform attention exactly like the reference's -finfo.max fill), causal mask
// j > Lk - Nq + i (right-aligned), optional in-kernel dropout on the probabilities
// (counter-hash RNG regenerated in the backward).
//
// Structure (v3):
//   workgroup = 4 waves x 64 lanes; each wave owns QH*16 q rows; KVBLK=64-key
//   K/V tiles cooperatively staged in LDS — K row-major (contiguous b128
//   fragment reads, +16 B row padding for conflict-free 16-lane groups), V as
//   16-column subtiles written row-major and consumed via ds_read_b64_tr_b16
//   hardware transpose reads; QK^T and PV on mfma_f32_16x16x32_bf16 with fp32
//   accumulation; online softmax (rescale-skip) with cross-lane shfl_xor row
//   reductions; P redistributed C-layout -> A-layout through a per-wave LDS
//   buffer; KV-split over grid.z with logsumexp merge. Outputs O + logsumexp.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <cfloat>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

namespace {

constexpr int KVBLK = 64;       // keys per tile
constexpr int KEYBLKS = KVBLK / 16;
constexpr int NWAVES = 4;       // waves per workgroup

DEVINL float warp16_max(float x) {
#pragma unroll
    for (int m = 1; m < 16; m <<= 1) x = fmaxf(x, __shfl_xor(x, m, 64));
    return x;
}

DEVINL float warp16_sum(float x) {
#pragma unroll
    for (int m = 1; m < 16; m <<= 1) x += __shfl_xor(x, m, 64);
    return x;
}

// Stage a (KVBLK x D) bf16 tile from global (row stride src_stride elems) into LDS
// row-major (row stride ldst_bytes), zero-padding col >= d / row >= rows_valid.
DEVINL void stage_tile_rowmajor(const unsigned short* __restrict__ src, long src_stride,
                                int rows_valid, int d, int d_pad,
                                char* lds, int ldst_bytes, int tid) {
    const int granules_per_row = d_pad / 8;
    const int total = KVBLK * granules_per_row;
    for (int g = tid; g < total; g += 256) {
        int row = g / granules_per_row;
        int c0 = (g % granules_per_row) * 8;
        short8v val = {};
        if (row < rows_valid) {
            if (c0 + 8 <= d) {
                val = *reinterpret_cast<const short8v*>(src + (long)row * src_stride + c0);
            } else {
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    val[e] = (c0 + e < d) ? (short)src[(long)row * src_stride + c0 + e] : (short)0;
                }
            }
        }
        *reinterpret_cast<short8v*>(lds + row * ldst_bytes + c0 * 2) = val;
    }
}

typedef __attribute__((ext_vector_type(4))) short short4x;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// ---- 16-column-subtiled image + hardware transpose read (guide T10) ----
// Layout: channel-subtile s holds a (KVBLK x 16) row-major bf16 block with
// 16-elem (32 B) rows, subtiles padded by 8 elems (16 B) so consecutive
// subtiles shift LDS banks. ds_read_b64_tr_b16's gather (lane l elem j reads
// base + (l&15) + j*16 + (l>>4)*64) turns two reads of this image into the
// 32x16 MFMA B-fragment — V is staged ROW-major (one 16-B write per granule,
// no 8-row write lockstep: that pattern measured ~11% of wave cycles in bank
// conflicts) yet consumed column-major for P@V.
constexpr int SUB_ELEMS = 64 * 16 + 8;  // KVBLK rows x 16 cols + 16-B pad

DEVINL void stage_tile_sub16(const unsigned short* __restrict__ src, long src_stride,
                             int rows_valid, int dv, int dv_pad,
                             char* lds, int tid) {
    const int gpr = dv_pad / 8;
    const int total = KVBLK * gpr;
    for (int g = tid; g < total; g += 256) {
        int row = g / gpr;
        int c0 = (g % gpr) * 8;
        short8v val = {};
        if (row < rows_valid && c0 < dv) {
            if (c0 + 8 <= dv) {
                val = *reinterpret_cast<const short8v*>(src + (long)row * src_stride + c0);
            } else {
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    val[e] = (c0 + e < dv) ? (short)src[(long)row * src_stride + c0 + e] : (short)0;
            }
        }
        *reinterpret_cast<short8v*>(
            lds + ((c0 / 16) * SUB_ELEMS + row * 16 + (c0 % 16)) * 2) = val;
    }
}

// B-fragment (rows key0..key0+31, cols 16*sub..) from the subtiled image.
// Per-lane addressing: each 16-lane group loads one 4x16 row-major tile
// linearly (lane supplies its own 8-B chunk at lo16*4 elems) and the
// instruction's fixed cross-lane exchange delivers column lo16 to the lane;
// group hi4 starts at row hi4*8, the second read covers rows +4.
DEVINL bf16x8 read_bfrag_tr16(const char* lds, int sub, int key0, int hi4, int lo16) {
    const __bf16* base = reinterpret_cast<const __bf16*>(lds) +
                         sub * SUB_ELEMS + (key0 + hi4 * 8) * 16 + lo16 * 4;
    auto p = (__attribute__((address_space(3))) bf16x4*)base;
    bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
    bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p + 16);  // +4 rows
    bf16x8 out;
#pragma unroll
    for (int e = 0; e < 4; ++e) { out[e] = lo[e]; out[e + 4] = hi[e]; }
    return out;
}

// QH: 16-row A-fragments per wave (2 doubles MFMA work per B-fragment LDS read;
// 1 for the large-D templates where the register budget is spent on O accumulators)
template <int DMAX, int DVMAX, int QH>
__launch_bounds__(256, DMAX <= 160 ? 2 : 1)  // 2 waves/SIMD where the register
__global__ void flash_fwd_kernel(            // budget allows (not the 352 template)
    const unsigned short* __restrict__ qp,  // (B,H,Nq,D) bf16, pre-scaled
    const unsigned short* __restrict__ kp,  // (B,H,Lk,D)
    const unsigned short* __restrict__ vp,  // (B,H,Lk,Dv)
    const bool* __restrict__ pad,           // (B,Lk) or null
    unsigned short* __restrict__ op,        // (B,H,Nq,Dv)
    float* __restrict__ lsep,               // (B,H,Nq)
    float* __restrict__ o_part,             // (S,B,H,Nq,Dv) fp32, when gridDim.z > 1
    float* __restrict__ lse_part,           // (S,B,H,Nq)
    long kv_chunk,                          // keys per split (multiple of KVBLK)
    long qsb, long qsh, long qsn,           // q strides (elements; last dim contiguous)
    long ksb, long ksh, long ksn,           // k strides
    long vsb, long vsh, long vsn,           // v strides
    long osb, long osh, long osn,           // out strides (merged-heads layout)
    int B, int H, int Nq, int Lk, int D, int Dv, int causal,
    float drop_p, unsigned long long drop_seed) {
    const int d_pad = (D + 31) & ~31;
    const int dv_pad = (Dv + 15) & ~15;
    const int d_blocks = d_pad / 32;       // QK^T k-steps
    const int dv_blocks = dv_pad / 16;     // O column blocks

    constexpr int QROWS = 16 * QH;
    constexpr int QBLK = QROWS * NWAVES;
    const int tid = threadIdx.x;
    const int wave = tid / 64;
    const int lane = tid % 64;
    const int lo16 = lane & 15;
    const int hi4 = lane >> 4;

    const int bh = blockIdx.y;
    const int b = bh / H;
    const int hh = bh % H;
    const int q0 = blockIdx.x * QBLK + wave * QROWS;

    const unsigned short* qbase = qp + (long)b * qsb + (long)hh * qsh;
    const unsigned short* kbase = kp + (long)b * ksb + (long)hh * ksh;
    const unsigned short* vbase = vp + (long)b * vsb + (long)hh * vsh;
    const bool* padrow = pad ? pad + (long)b * Lk : nullptr;

    // ---- dynamic LDS carve (guideline 17: 16-B aligned offsets) ----
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int k_stride = d_pad * 2 + 16;
    const int vt_stride = KVBLK * 2 + 16;
    char* k_lds = smem;                                   // KVBLK * k_stride
    char* v16_lds = k_lds + KVBLK * k_stride;             // (dv_pad/16) * SUB_ELEMS elems
    char* p_lds = v16_lds + (DVMAX / 16) * SUB_ELEMS * 2; // per wave: QROWS * vt_stride
    char* p_mine = p_lds + wave * QROWS * vt_stride;

    // ---- Q fragments: lane holds A[i=lo16][k = hi4*8 + e] per 32-wide k-block ----
    short8v q_frag[QH][DMAX / 32];
#pragma unroll
    for (int h = 0; h < QH; ++h) {
        int qi = q0 + h * 16 + lo16;
        bool valid = qi < Nq;
        int qclamp = valid ? qi : Nq - 1;
        const unsigned short* qrow = qbase + (long)qclamp * qsn;
#pragma unroll
        for (int kb = 0; kb < DMAX / 32; ++kb) {
            short8v val = {};
            if (kb < d_blocks && valid) {
                int c0 = kb * 32 + hi4 * 8;
                if (c0 + 8 <= D) {
                    val = *reinterpret_cast<const short8v*>(qrow + c0);
                } else {
#pragma unroll
                    for (int e = 0; e < 8; ++e) val[e] = (c0 + e < D) ? (short)qrow[c0 + e] : (short)0;
                }
            }
            q_frag[h][kb] = val;
        }
    }

    // ---- accumulators (C layout: lane holds rows h*16 + hi4*4+r, col lo16 + 16*cb) ----
    float4v o_acc[QH][DVMAX / 16];
    float m_run[QH][4], l_run[QH][4];
#pragma unroll
    for (int h = 0; h < QH; ++h) {
#pragma unroll
        for (int cb = 0; cb < DVMAX / 16; ++cb) o_acc[h][cb] = float4v{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int r = 0; r < 4; ++r) { m_run[h][r] = -INFINITY; l_run[h][r] = 0.f; }
    }

    // causal: q row qi may attend keys j <= Lk - Nq + qi
    int kv_end = Lk;
    if (causal) {
        int q_hi = blockIdx.x * QBLK + QBLK - 1;  // workgroup-max q row
        kv_end = min(Lk, Lk - Nq + q_hi + 1);
    }
    // KV-split across gridDim.z (fills the 256 CUs for small B*H*Nq/QBLK grids,
    // e.g. the 50k/182k-KV encoder cross-attentions and single-token decode);
    // partials merge in flash_merge_kernel via logsumexp weights.
    int kv_begin = 0;
    if (gridDim.z > 1) {
        kv_begin = (int)((long)blockIdx.z * kv_chunk);
        kv_end = min((long)kv_end, (long)(blockIdx.z + 1) * kv_chunk);
    }

    const unsigned int drop_thresh = (unsigned int)(drop_p * 65536.0);

    // T14 split staging on exact template matches (odd dims are padded to the
    // template by the host): K/V loads for tile t+1 fly under tile t's MFMAs
    // instead of serializing between the barriers — at the occupancy-1
    // large-D templates the synchronous staging was ~85% of the kernel
    constexpr bool kFast = (DMAX % 32 == 0) && (DVMAX % 32 == 0) && DMAX <= 288 &&
                           ((KVBLK * DMAX) % 2048 == 0) && ((KVBLK * DVMAX) % 2048 == 0);
    constexpr int NG_K = kFast ? (KVBLK * DMAX) / 2048 : 1;
    constexpr int NG_V = kFast ? (KVBLK * DVMAX) / 2048 : 1;
    short8v st_k[NG_K], st_v[NG_V];
    const bool fast = kFast && D == DMAX && Dv == DVMAX;
    const int n_tiles = (kv_end > kv_begin) ? (kv_end - kv_begin + KVBLK - 1) / KVBLK : 0;
    const int kv_last = kv_begin + (n_tiles > 0 ? (n_tiles - 1) * KVBLK : 0);
    if (fast && n_tiles > 0) {
        issue_tile<KVBLK, DMAX>(st_k, kbase + (long)kv_begin * ksn, ksn, Lk - kv_begin, tid);
        issue_tile<KVBLK, DVMAX>(st_v, vbase + (long)kv_begin * vsn, vsn, Lk - kv_begin, tid);
    }

    for (int kv0 = kv_begin; kv0 < kv_end; kv0 += KVBLK) {
        int rows_valid = min(KVBLK, Lk - kv0);
        __syncthreads();
        if (fast) {
            write_rm<KVBLK, DMAX>(st_k, k_lds, k_stride, rows_valid, tid);
            write_sub16<KVBLK, DVMAX>(st_v, v16_lds, rows_valid, tid);
            int kv_n = min(kv0 + KVBLK, kv_last);
            issue_tile<KVBLK, DMAX>(st_k, kbase + (long)kv_n * ksn, ksn, Lk - kv_n, tid);
            issue_tile<KVBLK, DVMAX>(st_v, vbase + (long)kv_n * vsn, vsn, Lk - kv_n, tid);
        } else {
            stage_tile_rowmajor(kbase + (long)kv0 * ksn, ksn, rows_valid, D, d_pad, k_lds, k_stride, tid);
            stage_tile_sub16(vbase + (long)kv0 * vsn, vsn, rows_valid, Dv, dv_pad, v16_lds, tid);
        }
        __syncthreads();

        // ---- S = Q K^T (QH x 16 rows x KVBLK keys); one B read feeds QH MFMAs ----
        float4v s_acc[QH][KEYBLKS];
#pragma unroll
        for (int h = 0; h < QH; ++h)
#pragma unroll
            for (int kb = 0; kb < KEYBLKS; ++kb) s_acc[h][kb] = float4v{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kb = 0; kb < DMAX / 32; ++kb) {
            if (kb < d_blocks) {
                // B fragment: B[k][j] = K[key j][ch k]; lane reads K row (lo16 + 16*keyblk)
#pragma unroll
                for (int keyblk = 0; keyblk < KEYBLKS; ++keyblk) {
                    const char* src = k_lds + (keyblk * 16 + lo16) * k_stride + (kb * 32 + hi4 * 8) * 2;
                    bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
#pragma unroll
                    for (int h = 0; h < QH; ++h) {
                        s_acc[h][keyblk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            (bf16x8)q_frag[h][kb], bfrag, s_acc[h][keyblk], 0, 0, 0);
                    }
                }
            }
        }

        // ---- mask + online softmax per 16-row group (P computed in place in
        // s_acc, dropped, and written straight to the per-wave LDS buffer).
        // Wave-uniform hoist: interior tiles of the common no-pad case need no
        // per-element masking at all (the mask chain is ~1/4 of the softmax
        // VALU work — the kernels measure VALU/wait-bound, profiles/) ----
        const bool tile_masked =
            (kv0 + KVBLK > Lk) || (padrow != nullptr) ||
            (causal && kv0 + KVBLK - 1 > Lk - Nq + q0);
#pragma unroll
        for (int h = 0; h < QH; ++h) {
            float rowmax[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int qi = q0 + h * 16 + hi4 * 4 + r;
                float mx = -FLT_MAX;
                if (tile_masked) {
#pragma unroll
                    for (int kb = 0; kb < KEYBLKS; ++kb) {
                        int j = kv0 + kb * 16 + lo16;
                        float sv = s_acc[h][kb][r];
                        bool masked = j >= Lk;
                        if (padrow && j < Lk) masked |= padrow[j];
                        if (causal && j > Lk - Nq + qi) masked = true;
                        sv = masked ? -FLT_MAX : sv;
                        s_acc[h][kb][r] = sv;
                        mx = fmaxf(mx, sv);
                    }
                } else {
#pragma unroll
                    for (int kb = 0; kb < KEYBLKS; ++kb) mx = fmaxf(mx, s_acc[h][kb][r]);
                }
                rowmax[r] = warp16_max(mx);
            }
            // rescale-skip: once the running max is established, most tiles do
            // not raise it — skip the O-wide rescale + m update wholesale (the
            // branch must be wave-uniform, hence __any over the wave's rows)
            bool need = false;
#pragma unroll
            for (int r = 0; r < 4; ++r) need |= rowmax[r] > m_run[h][r];
            if (__any(need)) {
                float4v alpha4;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    float m_new = fmaxf(m_run[h][r], rowmax[r]);
                    alpha4[r] = __expf(m_run[h][r] - m_new);
                    m_run[h][r] = m_new;
                    l_run[h][r] *= alpha4[r];
                }
#pragma unroll
                for (int cb = 0; cb < DVMAX / 16; ++cb) o_acc[h][cb] *= alpha4;  // v_pk_mul
            }
            {
                // kb-outer so the exp input and row sums run on packed fp32;
                // the P stores stay scalar (rows are vt_stride apart)
                const float4v mrun4 = {m_run[h][0], m_run[h][1], m_run[h][2], m_run[h][3]};
                float4v psum4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
                for (int kb = 0; kb < KEYBLKS; ++kb) {
                    float4v e4 = s_acc[h][kb] - mrun4;  // v_pk_add
                    float pv[4];
#pragma unroll
                    for (int r = 0; r < 4; ++r) pv[r] = __expf(e4[r]);
                    psum4 += float4v{pv[0], pv[1], pv[2], pv[3]};
                    if (drop_p > 0.f) {
                        int j = kv0 + kb * 16 + lo16;
                        int qb2 = (q0 + h * 16 + hi4 * 4) >> 1;
                        unsigned int hh[2] = {rng_hash(drop_seed, bh, qb2, j),
                                              rng_hash(drop_seed, bh, qb2 + 1, j)};
#pragma unroll
                        for (int r = 0; r < 4; ++r) {
                            unsigned int d = (r & 1) ? (hh[r >> 1] >> 16) : (hh[r >> 1] & 0xffffu);
                            if (d < drop_thresh) pv[r] = 0.f;
                        }
                    }
#pragma unroll
                    for (int r = 0; r < 4; ++r)
                        *reinterpret_cast<unsigned short*>(
                            p_mine + (h * 16 + hi4 * 4 + r) * vt_stride + (kb * 16 + lo16) * 2) =
                            f2bf(pv[r]);
                }
#pragma unroll
                for (int r = 0; r < 4; ++r) l_run[h][r] += warp16_sum(psum4[r]);
            }
        }
        // wave-local LDS ordering for the P roundtrip (lgkmcnt only)
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        bf16x8 p_frag[QH][KEYBLKS / 2];
#pragma unroll
        for (int h = 0; h < QH; ++h)
#pragma unroll
            for (int kb32 = 0; kb32 < KEYBLKS / 2; ++kb32) {
                p_frag[h][kb32] = (bf16x8)(*reinterpret_cast<const short8v*>(
                    p_mine + (h * 16 + lo16) * vt_stride + (kb32 * 32 + hi4 * 8) * 2));
            }

        // ---- O += P V : B[k=key][j=ch] via hardware transpose reads of the
        // row-major subtiled V image ----
#pragma unroll
        for (int cb = 0; cb < DVMAX / 16; ++cb) {
            if (cb < dv_blocks) {
#pragma unroll
                for (int kb32 = 0; kb32 < KEYBLKS / 2; ++kb32) {
                    bf16x8 bfrag = read_bfrag_tr16(v16_lds, cb, kb32 * 32, hi4, lo16);
#pragma unroll
                    for (int h = 0; h < QH; ++h) {
                        o_acc[h][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            p_frag[h][kb32], bfrag, o_acc[h][cb], 0, 0, 0);
                    }
                }
            }
        }
    }

    // ---- epilogue: O /= l (x dropout keep-rate); direct bf16 store, or fp32
    // partials (one slab per KV split) for the merge kernel ----
#pragma unroll
    for (int h = 0; h < QH; ++h)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int qi = q0 + h * 16 + hi4 * 4 + r;
            if (qi >= Nq) continue;
            float l_eff = l_run[h][r] * (drop_p > 0.f ? (1.0f - drop_p) : 1.0f);
            float inv_l = (l_eff > 0.f) ? 1.0f / l_eff : 0.f;
            float lse_v = (l_run[h][r] > 0.f) ? m_run[h][r] + logf(l_run[h][r]) : -1e30f;
            if (gridDim.z > 1) {
                long row = ((long)blockIdx.z * B * H + bh) * Nq + qi;
                float* orow = o_part + row * Dv;
#pragma unroll
                for (int cb = 0; cb < DVMAX / 16; ++cb) {
                    int c = cb * 16 + lo16;
                    if (cb < dv_blocks && c < Dv) orow[c] = o_acc[h][cb][r] * inv_l;
                }
                if (lo16 == 0) lse_part[row] = lse_v;
            } else {
                unsigned short* orow = op + (long)b * osb + (long)hh * osh + (long)qi * osn;
#pragma unroll
                for (int cb = 0; cb < DVMAX / 16; ++cb) {
                    int c = cb * 16 + lo16;
                    if (cb < dv_blocks && c < Dv) orow[c] = f2bf(o_acc[h][cb][r] * inv_l);
                }
                if (lo16 == 0) lsep[(long)bh * Nq + qi] = lse_v;
            }
        }
}

// Merge KV-split partials: one wave per (b,h,q) row.
//   w_i = exp(lse_i - max) ; O = sum w_i O_i / sum w_i ; lse = max + log(sum w_i)
__global__ void flash_merge_kernel(const float* __restrict__ o_part,
                                   const float* __restrict__ lse_part,
                                   unsigned short* __restrict__ op,
                                   float* __restrict__ lsep,
                                   long osb, long osh, long osn, int H, int Nq,
                                   long rows, int nsplit, int dv) {
    long row = (long)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
    int lane = threadIdx.x % 64;
    if (row >= rows) return;

    __shared__ float wbuf[4][32];  // per-wave weights (runtime-indexed arrays
    int wv_id = (int)(threadIdx.x / 64); // would spill to scratch — rule 20)

    float mx = -1e30f;
    for (int i = 0; i < nsplit; ++i) mx = fmaxf(mx, lse_part[(long)i * rows + row]);
    float wsum = 0.f;
    for (int i = lane; i < nsplit; i += 64) {
        float wv = __expf(lse_part[(long)i * rows + row] - mx);
        wbuf[wv_id][i] = wv;
    }
    __builtin_amdgcn_s_waitcnt(0);
    for (int i = 0; i < nsplit; ++i) wsum += wbuf[wv_id][i];
    float inv = (wsum > 0.f) ? 1.0f / wsum : 0.f;

    long b = row / ((long)H * Nq);
    long rem = row % ((long)H * Nq);
    unsigned short* orow = op + b * osb + (rem / Nq) * osh + (rem % Nq) * osn;
    for (int c = lane; c < dv; c += 64) {
        float acc = 0.f;
        for (int i = 0; i < nsplit; ++i)
            acc += wbuf[wv_id][i] * o_part[((long)i * rows + row) * dv + c];
        orow[c] = f2bf(acc * inv);
    }
    if (lane == 0) lsep[row] = mx + logf(fmaxf(wsum, 1e-37f));
}

void launch_merge(float* o_part_p, float* lse_part_p, torch::Tensor& out, torch::Tensor& lse,
                  long rows, int nsplit, int Dv) {
    auto stream = at::cuda::getCurrentCUDAStream();
    int wpb = 4;
    long blocks = (rows + wpb - 1) / wpb;
    hipLaunchKernelGGL(flash_merge_kernel, dim3(blocks), dim3(64 * wpb), 0, stream,
                       o_part_p, lse_part_p,
                       reinterpret_cast<unsigned short*>(out.data_ptr()),
                       lse.data_ptr<float>(),
                       out.stride(0), out.stride(1), out.stride(2),
                       (int)out.size(1), (int)out.size(2), rows, nsplit, Dv);
    HIP_CHECK_LAST();
}

template <int DMAX, int DVMAX, int QH>
void launch_flash_fwd(const torch::Tensor& q, const torch::Tensor& k, const torch::Tensor& v,
                      const c10::optional<torch::Tensor>& pad_mask, bool causal,
                      float drop_p, unsigned long long drop_seed,
                      torch::Tensor& out, torch::Tensor& lse) {
    constexpr int QROWS = 16 * QH;
    constexpr int QBLK = QROWS * NWAVES;
    int B = q.size(0), H = q.size(1), Nq = q.size(2), D = q.size(3);
    int Lk = k.size(2), Dv = v.size(3);
    const int d_pad = (D + 31) & ~31;
    int k_stride = d_pad * 2 + 16;
    int vt_stride = KVBLK * 2 + 16;
    size_t smem = (size_t)KVBLK * k_stride + (size_t)(DVMAX / 16) * SUB_ELEMS * 2 +
                  (size_t)NWAVES * QROWS * vt_stride;
    int gx = (Nq + QBLK - 1) / QBLK;
    int gy = B * H;
    // KV-split: aim for ~2 blocks per CU (512 workgroups on 256 CUs)
    int nsplit = 1;
    long kv_chunk = Lk;
    if ((long)gx * gy < 512 && Lk > 4 * KVBLK) {
        int want = (int)(512 / ((long)gx * gy)) + 1;
        int max_split = (Lk + 4 * KVBLK - 1) / (4 * KVBLK);  // >= 4 tiles per split
        nsplit = std::min({want, max_split, 32});
        long tiles = (Lk + KVBLK - 1) / KVBLK;
        long tiles_per = (tiles + nsplit - 1) / nsplit;
        kv_chunk = tiles_per * KVBLK;
        nsplit = (int)((Lk + kv_chunk - 1) / kv_chunk);
    }
    dim3 grid(gx, gy, nsplit);
    const bool* padp = nullptr;
    if (pad_mask.has_value() && pad_mask->defined()) {
        padp = pad_mask->data_ptr<bool>();
    }
    auto stream = at::cuda::getCurrentCUDAStream();
    torch::Tensor o_part, lse_part;
    float* o_part_p = nullptr;
    float* lse_part_p = nullptr;
    long rows = (long)B * H * Nq;
    if (nsplit > 1) {
        o_part = torch::empty({(long)nsplit, rows, (long)Dv}, q.options().dtype(torch::kFloat32));
        lse_part = torch::empty({(long)nsplit, rows}, q.options().dtype(torch::kFloat32));
        o_part_p = o_part.data_ptr<float>();
        lse_part_p = lse_part.data_ptr<float>();
    }
    hipLaunchKernelGGL((flash_fwd_kernel<DMAX, DVMAX, QH>), grid, dim3(256), smem, stream,
                       reinterpret_cast<const unsigned short*>(q.data_ptr()),
                       reinterpret_cast<const unsigned short*>(k.data_ptr()),
                       reinterpret_cast<const unsigned short*>(v.data_ptr()),
                       padp,
                       reinterpret_cast<unsigned short*>(out.data_ptr()),
                       lse.data_ptr<float>(), o_part_p, lse_part_p, kv_chunk,
                       q.stride(0), q.stride(1), q.stride(2),
                       k.stride(0), k.stride(1), k.stride(2),
                       v.stride(0), v.stride(1), v.stride(2),
                       out.stride(0), out.stride(1), out.stride(2),
                       B, H, Nq, Lk, D, Dv, (int)causal, drop_p, drop_seed);
    HIP_CHECK_LAST();
    if (nsplit > 1) {
        launch_merge(o_part_p, lse_part_p, out, lse, rows, nsplit, Dv);
    }
}

}  // namespace

bool flash_supported_impl(long d_qk, long d_v, long needs_dropout) {
    (void)needs_dropout;  // in-kernel counter-hash dropout
    return d_qk <= 352 && d_v <= 352;
}

// Deep-pipelined forward (flash_fwd_pipe.hip) for the regular head-dim regimes
bool flash_fwd_pipe_applicable(long D, long Dv, long Nq);
void flash_fwd_pipe_launch(const torch::Tensor& q, const torch::Tensor& k,
                           const torch::Tensor& v, const bool* padp, bool causal,
                           float drop_p, unsigned long long drop_seed,
                           torch::Tensor& out, torch::Tensor& lse,
                           float* o_part_p, float* lse_part_p, long kv_chunk, int nsplit);

namespace {
bool pipe_enabled() {
    static int v = [] {
        const char* e = getenv("PERCEIVER_NO_PIPE");
        return (e && e[0] == '1') ? 0 : 1;
    }();
    return v != 0;
}
}  // namespace

std::vector<torch::Tensor> flash_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                     c10::optional<torch::Tensor> pad_mask, bool causal,
                                     double dropout_p, int64_t seed) {
    TORCH_CHECK(q.is_cuda() && k.is_cuda() && v.is_cuda());
    TORCH_CHECK(q.scalar_type() == torch::kBFloat16, "flash_fwd: bf16 only");
    TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4);
    // last dim must be contiguous; batch/head/seq strides are free (head-transposed
    // views and preallocated KV-cache buffers pass through without copies)
    if (q.stride(3) != 1) q = q.contiguous();
    if (k.stride(3) != 1) k = k.contiguous();
    if (v.stride(3) != 1) v = v.contiguous();
    int D = q.size(3), Dv = v.size(3);
    TORCH_CHECK(flash_supported_impl(D, Dv, 0), "flash_fwd: unsupported head dims ", D, " ", Dv);

    // odd large head dims (the img/flow D=261 class) run as the 288 template
    // with zero-padded channels: exact for attention (zero K/Q columns add 0
    // to scores; zero V columns are sliced off), and the exact template match
    // unlocks the T14 fast staging path
    if (D > 160 && (D != 288 || Dv != 288) && D <= 288 && Dv <= 288) {
        namespace F = torch::nn::functional;
        auto q288 = F::pad(q, F::PadFuncOptions({0, 288 - D}));
        auto k288 = F::pad(k, F::PadFuncOptions({0, 288 - D}));
        auto v288 = F::pad(v, F::PadFuncOptions({0, 288 - Dv}));
        auto res = flash_fwd(q288, k288, v288, pad_mask, causal, dropout_p, seed);
        return {res[0].narrow(-1, 0, Dv), res[1]};
    }

    // out lives in merged-heads (B, N, H, Dv) memory and is returned as the
    // (B, H, N, Dv) permuted view: the module's transpose+reshape after
    // attention then costs nothing (it was a 42 MB copy per layer on MLM)
    auto out = torch::empty({q.size(0), q.size(2), q.size(1), (long)Dv}, q.options())
                   .permute({0, 2, 1, 3});
    auto lse = torch::empty({q.size(0), q.size(1), q.size(2)}, q.options().dtype(torch::kFloat32));
    if (q.numel() == 0 || k.numel() == 0) {
        // degenerate shapes fall back to the eager path upstream; just return zeros
        out.zero_();
        lse.zero_();
        return {out, lse};
    }

    c10::optional<torch::Tensor> pm;
    if (pad_mask.has_value() && pad_mask->defined()) {
        pm = pad_mask->contiguous();
    }

    float dp = (float)dropout_p;
    unsigned long long sd = (unsigned long long)seed;

    if (pipe_enabled() && flash_fwd_pipe_applicable(D, Dv, q.size(2))) {
        int B = q.size(0), H = q.size(1), Nq = q.size(2), Lk = k.size(2);
        constexpr int PIPE_QBLK = 128;
        constexpr int KVB = 64;
        long gx = (Nq + PIPE_QBLK - 1) / PIPE_QBLK;
        long gy = (long)B * H;
        int nsplit = 1;
        long kv_chunk = Lk;
        if (gx * gy < 512 && Lk > 4 * KVB) {
            int want = (int)(512 / (gx * gy)) + 1;
            int max_split = (Lk + 4 * KVB - 1) / (4 * KVB);
            nsplit = std::min({want, max_split, 32});
            long tiles = (Lk + KVB - 1) / KVB;
            long tiles_per = (tiles + nsplit - 1) / nsplit;
            kv_chunk = tiles_per * KVB;
            nsplit = (int)((Lk + kv_chunk - 1) / kv_chunk);
        }
        torch::Tensor o_part, lse_part;
        float* o_part_p = nullptr;
        float* lse_part_p = nullptr;
        long rows = (long)B * H * Nq;
        if (nsplit > 1) {
            o_part = torch::empty({(long)nsplit, rows, (long)Dv}, q.options().dtype(torch::kFloat32));
            lse_part = torch::empty({(long)nsplit, rows}, q.options().dtype(torch::kFloat32));
            o_part_p = o_part.data_ptr<float>();
            lse_part_p = lse_part.data_ptr<float>();
        }
        const bool* padp = (pm.has_value() && pm->defined()) ? pm->data_ptr<bool>() : nullptr;
        flash_fwd_pipe_launch(q, k, v, padp, causal, dp, sd, out, lse,
                              o_part_p, lse_part_p, kv_chunk, nsplit);
        if (nsplit > 1) launch_merge(o_part_p, lse_part_p, out, lse, rows, nsplit, Dv);
        return {out, lse};
    }

    if (D <= 32 && Dv <= 160)       launch_flash_fwd<32, 160, 2>(q, k, v, pm, causal, dp, sd, out, lse);
    else if (D <= 64 && Dv <= 64)   launch_flash_fwd<64, 64, 2>(q, k, v, pm, causal, dp, sd, out, lse);
    else if (D <= 128 && Dv <= 128) launch_flash_fwd<128, 128, 2>(q, k, v, pm, causal, dp, sd, out, lse);
    else if (D <= 160 && Dv <= 160) launch_flash_fwd<160, 160, 1>(q, k, v, pm, causal, dp, sd, out, lse);
    else if (D <= 288 && Dv <= 288) launch_flash_fwd<288, 288, 1>(q, k, v, pm, causal, dp, sd, out, lse);
    else                            launch_flash_fwd<352, 352, 1>(q, k, v, pm, causal, dp, sd, out, lse);

    return {out, lse};
}
