// Shared helpers for the perceiver_amd CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>

#define DEVINL __device__ __forceinline__

// wave64 is the CDNA scheduling quantum; hard-coded per the CDNA4 guide.
constexpr int WAVE = 64;

// vector types for wide loads (guideline 13: always vectorize bf16)
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(2))) float float2v;

// bf16 <-> f32 on raw bits (round-to-nearest-even); avoids the hip_bf16 C++ API.
DEVINL float bf2f(unsigned short b) {
    unsigned int u = ((unsigned int)b) << 16;
    float f;
    __builtin_memcpy(&f, &u, 4);
    return f;
}

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2_t;

// packed pair convert: ONE v_cvt_pk_bf16_f32 for two floats (the scalar
// form needs 2 converts + shift + or); used by the P/dS image packing
DEVINL unsigned int f2bf2(float a, float b) {
    float2v f = {a, b};
    bf16x2_t p = __builtin_convertvector(f, bf16x2_t);
    unsigned int u;
    __builtin_memcpy(&u, &p, 4);
    return u;
}

typedef __attribute__((ext_vector_type(2))) unsigned int uint2v;

// four floats -> one 8-B store (two v_cvt_pk + one ds_write_b64): the two
// 4-B stores per lane at 32-B stride were a 2-way LDS bank conflict on the
// packed P/dS images (PMC: 10-12% of wave cycles)
DEVINL uint2v f2bf4(float a, float b, float c, float d) {
    return uint2v{f2bf2(a, b), f2bf2(c, d)};
}

DEVINL unsigned short f2bf(float f) {
    // native single-instruction convert (v_cvt_pk_bf16_f32, RTNE on gfx950);
    // the manual bit-math round used to cost 3-4 VALU ops per convert and
    // f2bf sits in every epilogue and the softmax P-packing hot paths
    __bf16 b = (__bf16)f;
    unsigned short u;
    __builtin_memcpy(&u, &b, 2);
    return u;
}

// counter-based RNG for attention dropout: the same (seed, bh, i, j) always
// yields the same draw, so the backward regenerates the forward's mask
// exactly without storing it. 32-bit combine + murmur3-style finalizer: the
// original splitmix64 chain lowered to ~3x the VALU ops, and the hash runs
// once per attention element — it was a measurable share of the backward
// kernels' 21:1 VALU:MFMA instruction ratio (profiles/ PMC).
// Dropout draw mapping: one 32-bit hash yields TWO 16-bit draws for the
// qi-pair (qi>>1, qi&1) at a given key j; every kernel (fwd, dq, dkv) uses
// drop16(seed,bh,qi,j) so forward and backward regenerate identical masks.
// The q-major pairing halves the hash VALU in the three kernels whose inner
// r-loop walks consecutive qi (fwd pipe, fwd v3, dq — the compiler hoists
// the shared hash); dkv (ki-major inner loop) pays the same cost as before.
// Keep threshold is 16-bit: drop_p quantized to 1/65536.
DEVINL unsigned int rng_hash(unsigned long long seed, int bh, int i, int j) {
    unsigned int x = (unsigned int)seed + (unsigned int)(seed >> 32) * 0x9E3779B9u;
    x += (unsigned int)bh * 0x85EBCA6Bu + (unsigned int)i * 0xC2B2AE35u +
         (unsigned int)j * 0x27D4EB2Fu;
    x ^= x >> 16;
    x *= 0x7FEB352Du;
    x ^= x >> 15;
    x *= 0x846CA68Bu;
    x ^= x >> 16;
    return x;
}

DEVINL unsigned int drop16(unsigned long long seed, int bh, int qi, int j) {
    unsigned int h = rng_hash(seed, bh, qi >> 1, j);
    return (qi & 1) ? (h >> 16) : (h & 0xffffu);
}

// erf via the Abramowitz-Stegun 7.1.26 rational approximation (|err| <=
// 1.5e-7 absolute — below bf16 output resolution, so GELU results match the
// exact-erf torch.nn.GELU after rounding). libm's float-exact erff costs
// several times more VALU and made the fused GELU kernels compute-bound.
DEVINL float erf_fast(float x) {
    const float ax = fabsf(x);
    const float t = __frcp_rn(1.0f + 0.3275911f * ax);
    float poly = 1.061405429f;
    poly = poly * t - 1.453152027f;
    poly = poly * t + 1.421413741f;
    poly = poly * t - 0.284496736f;
    poly = poly * t + 0.254829592f;
    float e = 1.0f - poly * t * __expf(-ax * ax);
    return copysignf(e, x);
}

// GELU matching torch.nn.GELU default (erf form) to bf16 output precision
DEVINL float gelu_f(float x) { return 0.5f * x * (1.0f + erf_fast(x * 0.70710678118654752440f)); }

DEVINL float gelu_grad_f(float x) {
    const float kInvSqrt2 = 0.70710678118654752440f;
    const float kInvSqrt2Pi = 0.39894228040143267794f;
    float cdf = 0.5f * (1.0f + erf_fast(x * kInvSqrt2));
    float pdf = kInvSqrt2Pi * __expf(-0.5f * x * x);
    return cdf + x * pdf;
}

#define HIP_CHECK_LAST()                                                            \
    do {                                                                            \
        hipError_t e = hipGetLastError();                                           \
        if (e != hipSuccess) {                                                      \
            TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e)); \
        }                                                                           \
    } while (0)

// ---- T14 async-STAGE split helpers (shared by the attention kernels) ----
// issue: clamped-row unconditional vector loads into registers; the vmcnt
// waits land at the write call one compute phase later, hiding HBM latency
// under MFMA work. Valid only when the runtime dims match the template
// exactly (every granule full-width; callers pad odd dims to the template).
template <int ROWS_TILE, int DD, int NG>
DEVINL void issue_tile(short8v (&st)[NG], const unsigned short* __restrict__ src,
                       long sstride, int row_limit, int tid) {
    constexpr int GPR = DD / 8;
#pragma unroll
    for (int i = 0; i < NG; ++i) {
        int g = tid + i * 256;
        int row = g / GPR, c0 = (g % GPR) * 8;
        int rowc = row < row_limit ? row : row_limit - 1;
        st[i] = *reinterpret_cast<const short8v*>(src + (long)rowc * sstride + c0);
    }
}

// write halves: spill staged registers into the row-major and/or 16-column
// subtiled LDS images, zero-filling clamp-duplicated tail rows
template <int ROWS_TILE, int DD, int NG>
DEVINL void write_rm(const short8v (&st)[NG], char* lds_rm, int ldst_bytes,
                     int rows_valid, int tid) {
    constexpr int GPR = DD / 8;
    const bool tail = rows_valid < ROWS_TILE;
#pragma unroll
    for (int i = 0; i < NG; ++i) {
        int g = tid + i * 256;
        // NG rounds up for tiles that don't divide into whole 2048-elem
        // sweeps (288/352-wide pad tiers); the surplus granules are no-ops
        if ((ROWS_TILE * DD) % 2048 != 0 && g >= (ROWS_TILE * DD) / 8) continue;
        int row = g / GPR, c0 = (g % GPR) * 8;
        short8v val = st[i];
        if (tail && row >= rows_valid) val = short8v{};
        *reinterpret_cast<short8v*>(lds_rm + row * ldst_bytes + c0 * 2) = val;
    }
}

template <int ROWS_TILE, int DD, int NG>
DEVINL void write_sub16(const short8v (&st)[NG], char* lds16, int rows_valid, int tid) {
    constexpr int GPR = DD / 8;
    constexpr int SUBE = ROWS_TILE * 16 + 8;
    const bool tail = rows_valid < ROWS_TILE;
#pragma unroll
    for (int i = 0; i < NG; ++i) {
        int g = tid + i * 256;
        if ((ROWS_TILE * DD) % 2048 != 0 && g >= (ROWS_TILE * DD) / 8) continue;
        int row = g / GPR, c0 = (g % GPR) * 8;
        short8v val = st[i];
        if (tail && row >= rows_valid) val = short8v{};
        *reinterpret_cast<short8v*>(
            lds16 + ((c0 / 16) * SUBE + row * 16 + (c0 % 16)) * 2) = val;
    }
}

template <int ROWS_TILE, int DD, int NG>
DEVINL void write_rm_sub16_c(const short8v (&st)[NG], char* lds_rm, int ldst_bytes,
                             char* lds16, int rows_valid, int tid) {
    constexpr int GPR = DD / 8;
    constexpr int SUBE = ROWS_TILE * 16 + 8;
    const bool tail = rows_valid < ROWS_TILE;
#pragma unroll
    for (int i = 0; i < NG; ++i) {
        int g = tid + i * 256;
        if ((ROWS_TILE * DD) % 2048 != 0 && g >= (ROWS_TILE * DD) / 8) continue;
        int row = g / GPR, c0 = (g % GPR) * 8;
        short8v val = st[i];
        if (tail && row >= rows_valid) val = short8v{};
        *reinterpret_cast<short8v*>(lds_rm + row * ldst_bytes + c0 * 2) = val;
        *reinterpret_cast<short8v*>(
            lds16 + ((c0 / 16) * SUBE + row * 16 + (c0 % 16)) * 2) = val;
    }
}
