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

DEVINL unsigned short f2bf(float f) {
    unsigned int u;
    __builtin_memcpy(&u, &f, 4);
    unsigned int rounding = 0x7FFFu + ((u >> 16) & 1u);
    u += rounding;
    return (unsigned short)(u >> 16);
}

// counter-based RNG (splitmix64 finalizer) for attention dropout: the same
// (seed, bh, i, j) always yields the same draw, so the backward regenerates the
// forward's mask exactly without storing it.
DEVINL unsigned int rng_hash(unsigned long long seed, int bh, int i, int j) {
    unsigned long long z = seed + (unsigned long long)(unsigned)bh * 0x9E3779B97F4A7C15ull +
                           (unsigned long long)(unsigned)i * 0xBF58476D1CE4E5B9ull +
                           (unsigned long long)(unsigned)j * 0x94D049BB133111EBull;
    z ^= z >> 30; z *= 0xBF58476D1CE4E5B9ull;
    z ^= z >> 27; z *= 0x94D049BB133111EBull;
    z ^= z >> 31;
    return (unsigned int)z;
}

// exact-erf GELU, matches torch.nn.GELU default
DEVINL float gelu_f(float x) { return 0.5f * x * (1.0f + erff(x * 0.70710678118654752440f)); }

DEVINL float gelu_grad_f(float x) {
    const float kInvSqrt2 = 0.70710678118654752440f;
    const float kInvSqrt2Pi = 0.39894228040143267794f;
    float cdf = 0.5f * (1.0f + erff(x * kInvSqrt2));
    float pdf = kInvSqrt2Pi * expf(-0.5f * x * x);
    return cdf + x * pdf;
}

#define HIP_CHECK_LAST()                                                            \
    do {                                                                            \
        hipError_t e = hipGetLastError();                                           \
        if (e != hipSuccess) {                                                      \
            TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e)); \
        }                                                                           \
    } while (0)
