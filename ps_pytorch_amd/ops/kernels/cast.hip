// Vectorized wire pack/unpack (f32 <-> bf16) for gradient/weight transport.
//
// Role-equivalent of the reference's blosc-snappy pack/unpack
// (ref: src/compression.py:18-46): shrink bytes-on-wire before RCCL moves
// them over xGMI. The MI355X-native scheme is GPU-resident dtype truncation
// (f32 -> bf16, 2x) — the wire buffer never leaves HBM and the pack runs at
// HBM stream rate, where blosc ran on the host CPU behind a D2H copy.
//
// Arbitrary n: 4-wide vector main loop + scalar tail on thread 0 (buffers
// here are bucket slices at parameter boundaries, not always 16B-sized).
#include "common.h"

typedef unsigned short ushort8_t __attribute__((ext_vector_type(8)));

__global__ __launch_bounds__(256) void pack_bf16_kernel(
    unsigned short* __restrict__ dst, const float* __restrict__ src, long n)
{
    EW_IDX
    long nvec = n >> 2;
    for (long i = gid; i < nvec; i += stride) {
        float4_t v = ((const float4_t*)src)[i];
        ushort4_t o = {f32_to_bf16(v.x), f32_to_bf16(v.y), f32_to_bf16(v.z), f32_to_bf16(v.w)};
        ((unsigned long long*)dst)[i] = *(unsigned long long*)&o;
    }
    if (gid == 0)
        for (long i = nvec << 2; i < n; ++i) dst[i] = f32_to_bf16(src[i]);
}

__global__ __launch_bounds__(256) void unpack_bf16_kernel(
    float* __restrict__ dst, const unsigned short* __restrict__ src, long n)
{
    EW_IDX
    long nvec = n >> 2;
    for (long i = gid; i < nvec; i += stride) {
        unsigned long long u = ((const unsigned long long*)src)[i];
        ushort4_t s = *(ushort4_t*)&u;
        ((float4_t*)dst)[i] = {bf16_to_f32(s.x), bf16_to_f32(s.y), bf16_to_f32(s.z), bf16_to_f32(s.w)};
    }
    if (gid == 0)
        for (long i = nvec << 2; i < n; ++i) dst[i] = bf16_to_f32(src[i]);
}

extern "C" void ps_pack_bf16(void* dst, const void* src, long n, void* stream) {
    int blocks; ew_grid(n / 4, 256, &blocks);
    hipLaunchKernelGGL(pack_bf16_kernel, dim3(blocks), dim3(256), 0, (hipStream_t)stream,
                       (unsigned short*)dst, (const float*)src, n);
}

extern "C" void ps_unpack_bf16(void* dst, const void* src, long n, void* stream) {
    int blocks; ew_grid(n / 4, 256, &blocks);
    hipLaunchKernelGGL(unpack_bf16_kernel, dim3(blocks), dim3(256), 0, (hipStream_t)stream,
                       (float*)dst, (const unsigned short*)src, n);
}

// Channel pad for the C<=3 stem path: out[p][0:4] = {in[p][0:C], 0...}.
// With C=4, the SMALL implicit-GEMM gather becomes two aligned 8-B tap
// loads per 16-B quantum instead of 8 scalar bounds-checked gathers (the
// R50 7x7 stem fwd was ~6% of the step on the scalar form).
__global__ __launch_bounds__(256) void pad4_kernel(
    unsigned short* __restrict__ dst, const unsigned short* __restrict__ src,
    long npix, int C)
{
    EW_IDX
    for (long i = gid; i < npix; i += stride) {
        const unsigned short* s = src + i * C;
        ushort4_t o = {0, 0, 0, 0};
        for (int c = 0; c < C; ++c) o[c] = s[c];
        *(ushort4_t*)(dst + i * 4) = o;
    }
}

extern "C" void ps_pad4(void* dst, const void* src, long npix, int C,
                        void* stream) {
    int blocks; ew_grid(npix, 256, &blocks);
    hipLaunchKernelGGL(pad4_kernel, dim3(blocks), dim3(256), 0,
                       (hipStream_t)stream, (unsigned short*)dst,
                       (const unsigned short*)src, npix, C);
}

// general channel pad C -> CP (CP % 8 == 0 or % 8 == 4): out[p][0:CP] =
// {in[p][0:C], 0...}. One thread per OCTET, octet index fastest-varying so
// adjacent lanes write adjacent 16-B chunks of the same row (coalesced) —
// the old one-thread-per-ROW form ran the fc-sized pads (8192 rows x 512
// cols) on 8K threads looping 64 octets each: 1.13 ms/step on LeNet.
__global__ __launch_bounds__(256) void padc_kernel(
    unsigned short* __restrict__ dst, const unsigned short* __restrict__ src,
    long total, int C, int CP, int oct, unsigned long long moct)
{
    EW_IDX
    for (long i = gid; i < total; i += stride) {
        // fdiv_magic is undefined for d == 1 (single-octet rows: CP <= 8)
        unsigned int p = (oct == 1) ? (unsigned int)i
                                    : fdiv_u32((unsigned int)i, moct);
        int cb = ((unsigned int)i - p * oct) * 8;
        const unsigned short* s = src + (long)p * C + cb;
        unsigned short* d = dst + (long)p * CP + cb;
        int nc = C - cb;                 // valid elems in this octet (can be <=0)
        if (cb + 8 <= CP) {
            ushort8_t o = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
            for (int u = 0; u < 8; ++u)
                if (u < nc) o[u] = s[u];
            *(ushort8_t*)d = o;
        } else {                         // CP % 8 == 4 tail (the C<=3 stems)
            ushort4_t o = {0, 0, 0, 0};
#pragma unroll
            for (int u = 0; u < 4; ++u)
                if (u < nc) o[u] = s[u];
            *(ushort4_t*)d = o;
        }
    }
}

extern "C" void ps_padc(void* dst, const void* src, long npix, int C,
                        int CP, void* stream) {
    int oct = (CP + 7) >> 3;             // 16-B chunks per row (incl. 8-B tail)
    long total = npix * oct;             // fdiv_u32 is exact for 32-bit n —
                                         // all call sites are far below 2^32
    int blocks; ew_grid(total, 256, &blocks);
    hipLaunchKernelGGL(padc_kernel, dim3(blocks), dim3(256), 0,
                       (hipStream_t)stream, (unsigned short*)dst,
                       (const unsigned short*)src, total, C, CP, oct,
                       fdiv_magic(oct));
}
