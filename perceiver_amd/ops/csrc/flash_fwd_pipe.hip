// Deep-pipelined flash-attention forward for CDNA4 (gfx950) — round-2 rewrite of
// the v3 kernel for the regular head-dim regimes.
//
// What changed vs flash_fwd.hip (v3), per the round-1 PMC evidence (the v3
// kernels measure 12-18% MFMA-instruction share, wait/VALU-bound):
//   * T14 async-STAGE split: the K/V tile for step t+1 is issued as plain
//     global loads right after the LDS double-buffer flips, and its ds_write
//     pass runs one full compute phase later — HBM latency (~900 cy) hides
//     under the tile-t MFMAs instead of serializing between barriers.
//   * 1 block/CU register budget (__launch_bounds__(256, 1)): room for the O
//     accumulators (QH x Dv/16 float4), one S state, and the staged K/V
//     registers without the spills that killed this structure at 2 waves/SIMD.
//   * P roundtrip rebuilt: softmax writes P TRANSPOSED ([key][row] 16-col
//     subtile image) as packed 2-row ushort2 (v_cvt_pk) b32 stores — 16 packed
//     writes/wave/tile instead of 64 scalar b16 stores — and PV reads the A
//     fragment back with ds_read_b64_tr_b16 hardware transpose reads.
//   * pad/oob masking precomputed once per tile into a per-key f32 bias row
//     (staged with K), removing the per-element global pad loads and most of
//     the per-element mask VALU chain from the softmax hot path.
//   * epilogue stages O rows through LDS and stores dwordx4 rows (T21-style
//     widening) instead of 80 scalar 2-byte global stores per lane.
//
// Semantics are identical to flash_fwd.hip: q pre-scaled, pad_mask True=pad
// filled with -FLT_MAX, right-aligned causal (j > Lk - Nq + i masked), online
// softmax with rescale-skip, in-kernel counter-hash dropout, KV-split partials
// merged by the caller's flash_merge kernel. Reference semantics:
// /root/reference/perceiver/model/core/modules.py:130-166.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <cfloat>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) short short4x;

namespace {

constexpr int KVBLK = 64;        // keys per tile
constexpr int KEYBLKS = KVBLK / 16;
constexpr int NWAVES = 4;
constexpr int QH = 2;            // 16-row q fragments per wave
constexpr int QROWS = 16 * QH;   // 32 q rows per wave
constexpr int QBLK = QROWS * NWAVES;
constexpr int SUB_ELEMS = KVBLK * 16 + 8;  // 16-col subtile + 16-B bank shift

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

// B/A fragment via hardware transpose read of a [KVBLK][16] row-major image
// (V channel subtiles and the transposed-P image share this layout).
DEVINL bf16x8 read_frag_tr16(const char* lds, int sub, int key0, int hi4, int lo16) {
    const __bf16* base = reinterpret_cast<const __bf16*>(lds) +
                         sub * SUB_ELEMS + (key0 + hi4 * 8) * 16 + lo16 * 4;
    auto p = (__attribute__((address_space(3))) bf16x4*)base;
    bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
    bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p + 16);  // +4 keys
    bf16x8 out;
#pragma unroll
    for (int e = 0; e < 4; ++e) { out[e] = lo[e]; out[e + 4] = hi[e]; }
    return out;
}

// One K/V/bias staging step, register half (issue): clamped-row loads so the
// loads are unconditional (a per-element load guard would make hipcc branch
// around each load and drain vmcnt per element).
template <int KGR, int VGR, int DMAX, int DVMAX>
struct Staged {
    short8v k[KGR];
    short8v v[VGR];
    float bias;     // lane < KVBLK: -FLT_MAX where key padded/oob, else 0
};

template <int KGR, int VGR, int DMAX, int DVMAX>
DEVINL void stage_issue(Staged<KGR, VGR, DMAX, DVMAX>& st,
                        const unsigned short* __restrict__ kbase,
                        const unsigned short* __restrict__ vbase,
                        const bool* __restrict__ padrow,
                        long ksn, long vsn, int kv0, int Lk, int tid) {
    constexpr int GPR_K = DMAX / 8;
    constexpr int GPR_V = DVMAX / 8;
#pragma unroll
    for (int i = 0; i < KGR; ++i) {
        int g = tid + i * 256;
        int row = g / GPR_K;
        int c0 = (g % GPR_K) * 8;
        int rowc = min(kv0 + row, Lk - 1);
        st.k[i] = *reinterpret_cast<const short8v*>(kbase + (long)rowc * ksn + c0);
    }
#pragma unroll
    for (int i = 0; i < VGR; ++i) {
        int g = tid + i * 256;
        int row = g / GPR_V;
        int c0 = (g % GPR_V) * 8;
        int rowc = min(kv0 + row, Lk - 1);
        st.v[i] = *reinterpret_cast<const short8v*>(vbase + (long)rowc * vsn + c0);
    }
    if (tid < KVBLK) {
        int j = kv0 + tid;
        bool masked = j >= Lk;
        if (padrow != nullptr && j < Lk) masked |= padrow[j];
        st.bias = masked ? -FLT_MAX : 0.f;
    }
}

// Write half: ds_write the staged registers (compiler inserts the vmcnt waits
// here, one compute phase after the issue). K row-major (+16 B row pad), V and
// bias as their consumption layouts. Tail rows (>= Lk) were clamp-loaded; K
// needs no zeroing (S is masked via bias), V garbage is killed by P == 0
// except for fully-masked rows, where the reference's degrade-to-uniform
// contract needs V rows zeroed — tail tiles pay a small select chain.
template <int KGR, int VGR, int DMAX, int DVMAX>
DEVINL void stage_write(const Staged<KGR, VGR, DMAX, DVMAX>& st,
                        char* k_lds, char* v_lds, float* bias_lds,
                        int kv0, int Lk, int tid) {
    constexpr int GPR_K = DMAX / 8;
    constexpr int GPR_V = DVMAX / 8;
    constexpr int K_STRIDE = DMAX * 2 + 16;
    const bool tail = kv0 + KVBLK > Lk;
#pragma unroll
    for (int i = 0; i < KGR; ++i) {
        int g = tid + i * 256;
        int row = g / GPR_K;
        int c0 = (g % GPR_K) * 8;
        *reinterpret_cast<short8v*>(k_lds + row * K_STRIDE + c0 * 2) = st.k[i];
    }
#pragma unroll
    for (int i = 0; i < VGR; ++i) {
        int g = tid + i * 256;
        int row = g / GPR_V;
        int c0 = (g % GPR_V) * 8;
        short8v val = st.v[i];
        if (tail && kv0 + row >= Lk) val = short8v{};
        *reinterpret_cast<short8v*>(
            v_lds + ((c0 / 16) * SUB_ELEMS + row * 16 + (c0 % 16)) * 2) = val;
    }
    if (tid < KVBLK) bias_lds[tid] = st.bias;
}

// 2 blocks/CU for the small-D templates (LDS <= 80 KiB, register file tight
// but spill-free): the partner block's compute hides what ILP alone cannot at
// 1 block/CU. The D=128 templates keep the full register budget.
template <int DMAX, int DVMAX>
__launch_bounds__(256, (DMAX <= 64 && DVMAX <= 160 && !(DMAX == 64 && DVMAX == 128)) ? 2 : 1)
__global__ void flash_fwd_pipe_kernel(
    const unsigned short* __restrict__ qp,  // (B,H,Nq,D) bf16, pre-scaled
    const unsigned short* __restrict__ kp,
    const unsigned short* __restrict__ vp,
    const bool* __restrict__ pad,           // (B,Lk) or null
    unsigned short* __restrict__ op,        // (B,H,Nq,Dv)
    float* __restrict__ lsep,               // (B,H,Nq)
    float* __restrict__ o_part,             // (S,B,H,Nq,Dv) fp32 when gridDim.z>1
    float* __restrict__ lse_part,
    long kv_chunk,
    long qsb, long qsh, long qsn,
    long ksb, long ksh, long ksn,
    long vsb, long vsh, long vsn,
    long osb, long osh, long osn,  // out strides: flash returns the merged-
                                   // heads (B,N,H,Dv) memory layout as a view
    int B, int H, int Nq, int Lk, int causal,
    float drop_p, unsigned long long drop_seed) {
    constexpr int KGR = DMAX / 32;          // 16-B staging granules per thread
    constexpr int VGR = DVMAX / 32;
    constexpr int K_STRIDE = DMAX * 2 + 16;
    constexpr int NSUB = DVMAX / 16;
    constexpr int DBLOCKS = DMAX / 32;      // QK^T k-steps
    constexpr int CBLOCKS = DVMAX / 16;     // O column blocks

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

    // ---- LDS carve (one extern array; 16-B aligned offsets) ----
    extern __shared__ __attribute__((aligned(16))) char smem[];
    char* k_lds = smem;                                        // 2 x KVBLK x K_STRIDE
    char* v_lds = k_lds + 2 * KVBLK * K_STRIDE;                // 2 x NSUB x SUB_ELEMS x 2
    char* p_lds = v_lds + 2 * NSUB * SUB_ELEMS * 2;            // NWAVES x QH x SUB_ELEMS x 2
    float* bias_lds = reinterpret_cast<float*>(p_lds + NWAVES * QH * SUB_ELEMS * 2);  // 2 x KVBLK
    char* p_mine = p_lds + wave * QH * SUB_ELEMS * 2;

    // ---- Q fragments: lane holds A[i=lo16][k=hi4*8+e] per 32-wide k block ----
    short8v q_frag[QH][DBLOCKS];
#pragma unroll
    for (int h = 0; h < QH; ++h) {
        int qi = q0 + h * 16 + lo16;
        int qclamp = min(qi, Nq - 1);
        const unsigned short* qrow = qbase + (long)qclamp * qsn;
#pragma unroll
        for (int kb = 0; kb < DBLOCKS; ++kb) {
            q_frag[h][kb] = *reinterpret_cast<const short8v*>(qrow + kb * 32 + hi4 * 8);
        }
    }

    float4v o_acc[QH][CBLOCKS];
    float m_run[QH][4], l_run[QH][4];
#pragma unroll
    for (int h = 0; h < QH; ++h) {
#pragma unroll
        for (int cb = 0; cb < CBLOCKS; ++cb) o_acc[h][cb] = float4v{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int r = 0; r < 4; ++r) { m_run[h][r] = -INFINITY; l_run[h][r] = 0.f; }
    }

    int kv_end = Lk;
    if (causal) {
        int q_hi = blockIdx.x * QBLK + QBLK - 1;
        kv_end = min(Lk, Lk - Nq + q_hi + 1);
    }
    int kv_begin = 0;
    if (gridDim.z > 1) {
        kv_begin = (int)((long)blockIdx.z * kv_chunk);
        kv_end = min((long)kv_end, (long)(blockIdx.z + 1) * kv_chunk);
    }
    const int n_tiles = (kv_end - kv_begin + KVBLK - 1) / KVBLK;
    // prefetch clamp: issue addresses never run past the last real tile
    const int kv_last = kv_begin + (n_tiles > 0 ? (n_tiles - 1) * KVBLK : 0);

    const unsigned int drop_thresh = (unsigned int)(drop_p * 65536.0);

    Staged<KGR, VGR, DMAX, DVMAX> st;
    // ---- prologue: tile 0 staged synchronously, tile 1 issued ----
    if (n_tiles > 0) {
        stage_issue(st, kbase, vbase, padrow, ksn, vsn, kv_begin, Lk, tid);
        stage_write(st, k_lds, v_lds, bias_lds, kv_begin, Lk, tid);
        stage_issue(st, kbase, vbase, padrow, ksn, vsn,
                    min(kv_begin + KVBLK, kv_last), Lk, tid);
    }
    __syncthreads();

    // O += P(tile) @ V(tile) from the wave-private transposed-P image and the
    // given V buffer — called one tile LATE (T15-style single-tile software
    // pipeline) so these MFMAs sit beside the CURRENT tile's softmax VALU in
    // the instruction stream, hiding it on the separate pipes
    auto pv_accumulate = [&](const char* v_buf) {
#pragma unroll
        for (int kb32 = 0; kb32 < KEYBLKS / 2; ++kb32) {
            bf16x8 a_frag[QH];
#pragma unroll
            for (int h = 0; h < QH; ++h)
                a_frag[h] = read_frag_tr16(p_mine, h, kb32 * 32, hi4, lo16);
#pragma unroll
            for (int cb = 0; cb < CBLOCKS; ++cb) {
                bf16x8 bfrag = read_frag_tr16(v_buf, cb, kb32 * 32, hi4, lo16);
#pragma unroll
                for (int h = 0; h < QH; ++h) {
                    o_acc[h][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[h], bfrag, o_acc[h][cb], 0, 0, 0);
                }
            }
        }
    };

    for (int ti = 0; ti < n_tiles; ++ti) {
        const int kv0 = kv_begin + ti * KVBLK;
        const int buf = ti & 1;
        char* k_cur = k_lds + buf * KVBLK * K_STRIDE;
        const float* bias_cur = bias_lds + buf * KVBLK;

        // ---- S = Q K^T ----
        float4v s_acc[QH][KEYBLKS];
#pragma unroll
        for (int h = 0; h < QH; ++h)
#pragma unroll
            for (int kb = 0; kb < KEYBLKS; ++kb) s_acc[h][kb] = float4v{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kb = 0; kb < DBLOCKS; ++kb) {
#pragma unroll
            for (int keyblk = 0; keyblk < KEYBLKS; ++keyblk) {
                const char* src = k_cur + (keyblk * 16 + lo16) * K_STRIDE + (kb * 32 + hi4 * 8) * 2;
                bf16x8 bfrag = (bf16x8)(*reinterpret_cast<const short8v*>(src));
#pragma unroll
                for (int h = 0; h < QH; ++h) {
                    s_acc[h][keyblk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        (bf16x8)q_frag[h][kb], bfrag, s_acc[h][keyblk], 0, 0, 0);
                }
            }
        }

        // ---- previous tile's PV: independent of s_acc, so the scheduler
        // interleaves its MFMAs with the softmax VALU below. The rescale in
        // the softmax touches o_acc only after these MFMAs in program order —
        // exactly the T13-safe ordering (P(t-1) fully applied at its own
        // scale before tile t's max can rescale O). p_mine is rewritten only
        // below the reads (same wave; LDS ops complete in order). ----
        if (ti > 0) {
            pv_accumulate(v_lds + ((ti - 1) & 1) * NSUB * SUB_ELEMS * 2);
        }

        // ---- mask + online softmax; P packed transposed into LDS ----
        // per-key bias covers oob + pad; the causal chain only runs on tiles
        // the right-aligned mask actually cuts (wave-uniform hoist)
        float kbias[KEYBLKS];
#pragma unroll
        for (int kb = 0; kb < KEYBLKS; ++kb) kbias[kb] = bias_cur[kb * 16 + lo16];
        const bool causal_tile = causal && (kv0 + KVBLK - 1 > Lk - Nq + q0);

#pragma unroll
        for (int h = 0; h < QH; ++h) {
            float rowmax[4];
            // bias add + running max on packed fp32 (v_pk_add_f32 halves the
            // VALU); the causal mask stays a per-element select on the (rare,
            // wave-uniform-hoisted) cut tiles only
            float4v mx4 = {-FLT_MAX, -FLT_MAX, -FLT_MAX, -FLT_MAX};
#pragma unroll
            for (int kb = 0; kb < KEYBLKS; ++kb) {
                float4v s4 = s_acc[h][kb] + kbias[kb];
                if (causal_tile) {
                    int j = kv0 + kb * 16 + lo16;
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int jmax = Lk - Nq + (q0 + h * 16 + hi4 * 4 + r);
                        if (j > jmax) s4[r] = -FLT_MAX;
                    }
                }
                s_acc[h][kb] = s4;
                mx4 = __builtin_elementwise_max(mx4, s4);
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) rowmax[r] = warp16_max(mx4[r]);
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
                for (int cb = 0; cb < CBLOCKS; ++cb) o_acc[h][cb] *= alpha4;  // v_pk_mul
            }
            // exp + dropout + packed transposed write: rows (hi4*4+r, +r+1) are
            // adjacent elements of the [key][row] image -> one ushort2 store.
            // Row sums accumulate pre-dropout (the epilogue divides by
            // l * (1 - p), matching the backward's regenerated mask).
            float4v psum4 = {0.f, 0.f, 0.f, 0.f};
            const float4v mrun4 = {m_run[h][0], m_run[h][1], m_run[h][2], m_run[h][3]};
#pragma unroll
            for (int kb = 0; kb < KEYBLKS; ++kb) {
                float4v e4 = s_acc[h][kb] - mrun4;  // v_pk_add
                float pv[4];
#pragma unroll
                for (int r = 0; r < 4; ++r) pv[r] = __expf(e4[r]);
                psum4 += float4v{pv[0], pv[1], pv[2], pv[3]};
                if (drop_p > 0.f) {
                    // explicit qi-pair hashes (qi base is a multiple of 4, so
                    // rows r=0,1 and r=2,3 share one 32-bit hash each)
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
                unsigned short* dst = reinterpret_cast<unsigned short*>(
                    p_mine + (h * SUB_ELEMS + (kb * 16 + lo16) * 16 + hi4 * 4) * 2);
                *reinterpret_cast<uint2v*>(dst) = f2bf4(pv[0], pv[1], pv[2], pv[3]);
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) l_run[h][r] += warp16_sum(psum4[r]);
        }

        __syncthreads();  // B1: all waves done with tile ti's LDS buffers
        if (ti + 1 < n_tiles) {
            stage_write(st, k_lds + ((ti + 1) & 1) * KVBLK * K_STRIDE,
                        v_lds + ((ti + 1) & 1) * NSUB * SUB_ELEMS * 2,
                        bias_lds + ((ti + 1) & 1) * KVBLK,
                        kv0 + KVBLK, Lk, tid);
            stage_issue(st, kbase, vbase, padrow, ksn, vsn,
                        min(kv0 + 2 * KVBLK, kv_last), Lk, tid);
            __syncthreads();  // B2: tile ti+1 visible
        }
    }

    // drain the software pipeline: the last tile's PV
    if (n_tiles > 0) {
        pv_accumulate(v_lds + ((n_tiles - 1) & 1) * NSUB * SUB_ELEMS * 2);
    }

    // ---- epilogue: O /= l, staged through LDS, stored as dwordx4 rows ----
    __syncthreads();  // LDS free for reuse
    constexpr int OROW = DVMAX * 2;           // bytes per staged O row
    char* o_mine = smem + wave * QROWS * OROW;
    const float keep = (drop_p > 0.f) ? (1.0f - drop_p) : 1.0f;
#pragma unroll
    for (int h = 0; h < QH; ++h)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            float l_eff = l_run[h][r] * keep;
            float inv_l = (l_eff > 0.f) ? 1.0f / l_eff : 0.f;
#pragma unroll
            for (int cb = 0; cb < CBLOCKS; ++cb) {
                *reinterpret_cast<unsigned short*>(
                    o_mine + (h * 16 + hi4 * 4 + r) * OROW + (cb * 16 + lo16) * 2) =
                    f2bf(o_acc[h][cb][r] * inv_l);
            }
        }
    // lse while the LDS writes drain
    if (lo16 == 0) {
#pragma unroll
        for (int h = 0; h < QH; ++h)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int qi = q0 + h * 16 + hi4 * 4 + r;
                if (qi >= Nq) continue;
                float lse_v = (l_run[h][r] > 0.f) ? m_run[h][r] + logf(l_run[h][r]) : -1e30f;
                if (gridDim.z > 1)
                    lse_part[((long)blockIdx.z * B * H + bh) * Nq + qi] = lse_v;
                else
                    lsep[(long)bh * Nq + qi] = lse_v;
            }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    if (gridDim.z > 1) {
        // split path: fp32 partials, scalar stores (rare: tiny grids only)
        const char* orow_lds = o_mine;
#pragma unroll
        for (int h = 0; h < QH; ++h)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int qi = q0 + h * 16 + hi4 * 4 + r;
                if (qi >= Nq) continue;
                long row = ((long)blockIdx.z * B * H + bh) * Nq + qi;
                float* od = o_part + row * DVMAX;  // Dv == DVMAX enforced by launcher
#pragma unroll
                for (int cb = 0; cb < CBLOCKS; ++cb) {
                    int c = cb * 16 + lo16;
                    od[c] = bf2f(*reinterpret_cast<const unsigned short*>(
                        orow_lds + (h * 16 + hi4 * 4 + r) * OROW + c * 2));
                }
            }
    } else {
        constexpr int GPR_O = DVMAX / 8;      // 16-B granules per row
        constexpr int TOT = QROWS * GPR_O;
        const bool all_valid = q0 + QROWS <= Nq;
#pragma unroll
        for (int i = 0; i < (TOT + 63) / 64; ++i) {
            int g = lane + i * 64;
            int row = g / GPR_O;
            int c0 = (g % GPR_O) * 8;
            int qi = q0 + row;
            short8v val = *reinterpret_cast<const short8v*>(o_mine + row * OROW + c0 * 2);
            if (all_valid || qi < Nq) {
                *reinterpret_cast<short8v*>(
                    op + (long)b * osb + (long)hh * osh + (long)qi * osn + c0) = val;
            }
        }
    }
}

}  // namespace

// The packed-P / bias / dwordx4-epilogue layouts assume EXACT template dims
// (the kernel reads/writes DVMAX columns unconditionally), so only the
// instantiated (D, Dv) pairs qualify; anything else falls back to the v3
// kernel. Nq >= QBLK keeps every wave's q fragment populated.
bool flash_fwd_pipe_applicable(long D, long Dv, long Nq) {
    if (Nq < QBLK) return false;
    if (D == 32) return Dv == 32 || Dv == 64 || Dv == 96 || Dv == 128 || Dv == 160;
    if (D == 64) return Dv == 64 || Dv == 128;
    if (D == 128) return Dv == 128;
    return false;
}

namespace {
template <int DMAX, int DVMAX>
void launch_pipe(const torch::Tensor& q, const torch::Tensor& k, const torch::Tensor& v,
                 const bool* padp, bool causal, float drop_p, unsigned long long drop_seed,
                 torch::Tensor& out, torch::Tensor& lse,
                 float* o_part_p, float* lse_part_p, long kv_chunk, int nsplit) {
    int B = q.size(0), H = q.size(1), Nq = q.size(2);
    int Lk = k.size(2);
    constexpr int K_STRIDE = DMAX * 2 + 16;
    constexpr int NSUB = DVMAX / 16;
    size_t smem = 2 * KVBLK * K_STRIDE + 2 * NSUB * SUB_ELEMS * 2 +
                  NWAVES * QH * SUB_ELEMS * 2 + 2 * KVBLK * sizeof(float);
    size_t smem_epi = (size_t)NWAVES * QROWS * DVMAX * 2;
    if (smem_epi > smem) smem = smem_epi;
    if (smem > 65536) {
        // dynamic-LDS requests above the 64 KiB default need the opt-in
        // (the CU has 160 KiB; we run 1 block/CU)
        static bool raised = [] {
            hipFuncSetAttribute(
                reinterpret_cast<const void*>(&flash_fwd_pipe_kernel<DMAX, DVMAX>),
                hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
            return true;
        }();
        (void)raised;
    }
    int gx = (Nq + QBLK - 1) / QBLK;
    dim3 grid(gx, B * H, nsplit);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL((flash_fwd_pipe_kernel<DMAX, DVMAX>), grid, dim3(256), smem, stream,
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
                       B, H, Nq, Lk, (int)causal, drop_p, drop_seed);
    HIP_CHECK_LAST();
}
}  // namespace

// Launch the pipelined forward. Caller guarantees flash_fwd_pipe_applicable()
// and provides the split-partial buffers when nsplit > 1 (merged by the
// caller's flash_merge kernel, same contract as the v3 kernel).
void flash_fwd_pipe_launch(const torch::Tensor& q, const torch::Tensor& k,
                           const torch::Tensor& v, const bool* padp, bool causal,
                           float drop_p, unsigned long long drop_seed,
                           torch::Tensor& out, torch::Tensor& lse,
                           float* o_part_p, float* lse_part_p, long kv_chunk, int nsplit) {
    int D = q.size(3), Dv = v.size(3);
    if (D == 32 && Dv == 32)        launch_pipe<32, 32>(q, k, v, padp, causal, drop_p, drop_seed, out, lse, o_part_p, lse_part_p, kv_chunk, nsplit);
    else if (D == 32 && Dv == 64)   launch_pipe<32, 64>(q, k, v, padp, causal, drop_p, drop_seed, out, lse, o_part_p, lse_part_p, kv_chunk, nsplit);
    else if (D == 32 && Dv == 96)   launch_pipe<32, 96>(q, k, v, padp, causal, drop_p, drop_seed, out, lse, o_part_p, lse_part_p, kv_chunk, nsplit);
    else if (D == 32 && Dv == 128)  launch_pipe<32, 128>(q, k, v, padp, causal, drop_p, drop_seed, out, lse, o_part_p, lse_part_p, kv_chunk, nsplit);
    else if (D == 32 && Dv == 160)  launch_pipe<32, 160>(q, k, v, padp, causal, drop_p, drop_seed, out, lse, o_part_p, lse_part_p, kv_chunk, nsplit);
    else if (D == 64 && Dv == 64)   launch_pipe<64, 64>(q, k, v, padp, causal, drop_p, drop_seed, out, lse, o_part_p, lse_part_p, kv_chunk, nsplit);
    else if (D == 64 && Dv == 128)  launch_pipe<64, 128>(q, k, v, padp, causal, drop_p, drop_seed, out, lse, o_part_p, lse_part_p, kv_chunk, nsplit);
    else if (D == 128 && Dv == 128) launch_pipe<128, 128>(q, k, v, padp, causal, drop_p, drop_seed, out, lse, o_part_p, lse_part_p, kv_chunk, nsplit);
    else TORCH_CHECK(false, "flash_fwd_pipe: no template for D=", D, " Dv=", Dv);
}
