// Common device helpers for the flink_ms_amd CDNA4 (gfx950) kernels.
// Single HIP path, MI355X-only: wave64, MFMA bf16 16x16x32 tiles, LDS staging.
#pragma once

#include <hip/hip_fp8.h>
#include <hip/hip_runtime.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// MFMA fragment/accumulator register types (cdna_hip_programming.md §3):
// 8 bf16 (4 VGPRs) per A/B fragment, 4 fp32 accumulators per 16x16 C/D tile.
// fp8 e4m3 fragments are 8 bytes (2 VGPRs): the i64 operand of
// v_mfma_f32_16x16x32_fp8_fp8 (OCP e4m3fn on gfx950, NOT MI300X fnuz).
typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef long long fp8x8;

// float <-> OCP e4m3fn scalar conversions (round-to-nearest-even, matching
// torch.float8_e4m3fn casts; conversions sit in staging/epilogue, not the
// MFMA hot loop).
DEV_INLINE unsigned char f2fp8(float f) {
    __hip_fp8_e4m3 q(f);
    return q.__x;
}

DEV_INLINE float fp82f(unsigned char u) {
    __hip_fp8_e4m3 q;
    q.__x = u;
    return (float)q;
}

DEV_INLINE float bf2f(unsigned short u) {
    union { unsigned int i; float f; } v;
    v.i = ((unsigned int)u) << 16;
    return v.f;
}

DEV_INLINE unsigned short f2bf(float f) {
    union { float f; unsigned int i; } v;
    v.f = f;
    // round-to-nearest-even (matches torch bfloat16 casts for finite values)
    unsigned int r = v.i + 0x7fffu + ((v.i >> 16) & 1u);
    return (unsigned short)(r >> 16);
}

DEV_INLINE float wave_reduce_sum(float x) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        x += __shfl_xor(x, off, WAVE);
    return x;
}

// Upper-triangle 16x16-tile enumeration for the Gramian accumulator:
// tiles 0..KT*(KT+1)/2-1 cover (p,q) with p<=q row by row; the KT tiles after
// that are the (p, EXT) rating-column tiles that produce b.
constexpr int up_tile_p(int t, int KT) {
    int p = 0;
    while (t >= KT - p) { t -= KT - p; ++p; }
    return p;
}
constexpr int up_tile_q(int t, int KT) {
    int p = 0;
    while (t >= KT - p) { t -= KT - p; ++p; }
    return p + t;
}
