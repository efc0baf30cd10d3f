// Common helpers for ps_pytorch_amd CDNA4 (gfx950) kernels.
// Target: MI355X only — wave64, 256 CUs, 8 XCDs, HBM3E ~8 TB/s.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE 64

// dtype tags shared with the Python side (ps_pytorch_amd/ops/__init__.py)
enum PsDtype : int { PS_F32 = 0, PS_BF16 = 1 };

typedef float float4_t __attribute__((ext_vector_type(4)));
typedef unsigned short ushort4_t __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
    union { unsigned int i; float f; } v;
    v.i = ((unsigned int)u) << 16;
    return v.f;
}

// round-to-nearest-even f32 -> bf16
__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
    union { float f; unsigned int i; } v;
    v.f = f;
    unsigned int x = v.i;
    unsigned int lsb = (x >> 16) & 1u;
    x += 0x7fffu + lsb;
    // NaN stays NaN (quiet)
    if ((v.i & 0x7f800000u) == 0x7f800000u && (v.i & 0x007fffffu)) x = v.i | 0x00400000u;
    return (unsigned short)(x >> 16);
}

// Memory-bound elementwise launch geometry (guide §6 G11): cap the grid at
// ~8 blocks/CU and grid-stride the rest. 256 threads/block (4 waves).
static inline void ew_grid(long n_vec, int threads, int* blocks) {
    long want = (n_vec + threads - 1) / threads;
    long cap = 256L * 8L;
    *blocks = (int)(want < cap ? (want > 0 ? want : 1) : cap);
}

// Lemire fast division: q = floor(n/d) = umul64hi(n, floor(2^64/d)+1),
// exact for all 32-bit n and d > 1 (d == 1 encoded as magic 0). Replaces
// per-element integer division in pixel/channel decode loops.
__device__ __forceinline__ unsigned fdiv_u32(unsigned n,
                                             unsigned long long magic) {
    return magic ? (unsigned)__umul64hi((unsigned long long)n, magic) : n;
}
static inline unsigned long long fdiv_magic(long d) {
    return d > 1 ? (~0ULL) / (unsigned long long)d + 1 : 0ULL;
}

// Tail convention: kernels vectorize 4 elements/lane; the final n%4 scalar
// elements are handled by lane (gid == 0) of each kernel with scalar code.
#define EW_IDX long gid = (long)blockIdx.x * blockDim.x + threadIdx.x; \
               long stride = (long)gridDim.x * blockDim.x;
