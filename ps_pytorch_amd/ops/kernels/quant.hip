// Block-scaled int8 gradient compression + fan-in accumulate kernels.
//
// Role-equivalent of the reference's blosc-snappy gradient codec
// (ref: src/compression.py:18-46 g_compress/g_decompress): shrink
// bytes-on-wire before RCCL moves them over xGMI. blosc is a byte-oriented
// host codec behind a D2H copy; the MI355X-native scheme is GPU-resident
// fixed-rate quantization — 4x smaller than f32 (2x smaller than the bf16
// wire), deterministic payload size (RCCL needs sized buffers), and the
// pack runs at HBM stream rate.
//
// Scheme: 256-element blocks; per-block scale = max|x| / 127; values are
// int8 = rint(x * 127 / max|x|). Payload layout (one contiguous byte
// buffer so a bucket is ONE send):
//   [ int8 quants, n bytes, zero-padded to 4B ][ f32 scales, ceil(n/256) ]
//
// pack: one wave (64 lanes x 4 elems) per 256-block; wave max-reduce via
// XOR shuffles (no LDS), 4 waves per workgroup.
// unpack/unpack_acc: grid-stride, 4 elems/lane (a 4-aligned quad never
// straddles a 256-block).
#include "common.h"

typedef char char4_t __attribute__((ext_vector_type(4)));

#define Q8_BLOCK 256

static inline long q8_qbytes(long n) { return (n + 3) & ~3L; }
static inline long q8_nblk(long n) { return (n + Q8_BLOCK - 1) / Q8_BLOCK; }

template <bool BF16>
__global__ __launch_bounds__(256) void pack_q8_kernel(
    char* __restrict__ q, float* __restrict__ scales,
    const void* __restrict__ src_, long n, long nblk)
{
    const float* __restrict__ srcf = (const float*)src_;
    const unsigned short* __restrict__ srcb = (const unsigned short*)src_;
    long blk = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
    long stride = (long)gridDim.x * 4;
    for (; blk < nblk; blk += stride) {
        int lane = threadIdx.x & 63;
        long idx = blk * Q8_BLOCK + (long)lane * 4;
        float v[4];
        if (idx + 4 <= n) {
            if constexpr (BF16) {
                unsigned long long u = *(const unsigned long long*)(srcb + idx);
                ushort4_t s = *(ushort4_t*)&u;
                v[0] = bf16_to_f32(s.x); v[1] = bf16_to_f32(s.y);
                v[2] = bf16_to_f32(s.z); v[3] = bf16_to_f32(s.w);
            } else {
                float4_t x = *(const float4_t*)(srcf + idx);
                v[0] = x.x; v[1] = x.y; v[2] = x.z; v[3] = x.w;
            }
        } else {
#pragma unroll
            for (int k = 0; k < 4; ++k)
                v[k] = (idx + k < n)
                    ? (BF16 ? bf16_to_f32(srcb[idx + k]) : srcf[idx + k]) : 0.f;
        }
        float m = 0.f;
#pragma unroll
        for (int k = 0; k < 4; ++k) m = fmaxf(m, fabsf(v[k]));
#pragma unroll
        for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
        float inv = m > 0.f ? 127.f / m : 0.f;
        if (lane == 0) scales[blk] = m / 127.f;
        char4_t o;
#pragma unroll
        for (int k = 0; k < 4; ++k) {
            float r = rintf(v[k] * inv);
            r = fminf(127.f, fmaxf(-127.f, r));
            o[k] = (char)r;
        }
        if (idx + 4 <= n) {
            *(char4_t*)(q + idx) = o;
        } else {
            for (int k = 0; k < 4 && idx + k < n; ++k) q[idx + k] = o[k];
        }
    }
}

template <bool ACC>
__global__ __launch_bounds__(256) void unpack_q8_kernel(
    float* __restrict__ dst, const char* __restrict__ q,
    const float* __restrict__ scales, long n)
{
    EW_IDX
    long nvec = n >> 2;
    for (long i = gid; i < nvec; i += stride) {
        char4_t c = ((const char4_t*)q)[i];
        float s = scales[(i * 4) >> 8];   // quad never straddles a block
        float4_t d;
        if constexpr (ACC) d = ((const float4_t*)dst)[i];
        else               d = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int k = 0; k < 4; ++k) d[k] = fmaf((float)c[k], s, d[k]);
        ((float4_t*)dst)[i] = d;
    }
    if (gid == 0)
        for (long i = nvec << 2; i < n; ++i) {
            float val = (float)q[i] * scales[i >> 8];
            dst[i] = ACC ? dst[i] + val : val;
        }
}

// acc += src (wire-dtype fan-in accumulate for uncompressed gather mode:
// the PS sums per-worker staging buffers as they arrive, in arrival order —
// the reference's Waitany-drain += loop, sync_replicas_master_nn.py:157-186,
// run on-device).
template <typename SV>
__global__ __launch_bounds__(256) void acc_kernel(
    float* __restrict__ acc, const void* __restrict__ src_, long n)
{
    const SV* __restrict__ src = (const SV*)src_;
    EW_IDX
    long nvec = n >> 2;
    for (long i = gid; i < nvec; i += stride) {
        float4_t a = ((const float4_t*)acc)[i];
        SV sv = src[i];
        float4_t g;
        if constexpr (sizeof(SV) == 8) {
            ushort4_t u = *(ushort4_t*)&sv;
            g = {bf16_to_f32(u.x), bf16_to_f32(u.y), bf16_to_f32(u.z), bf16_to_f32(u.w)};
        } else {
            g = *(float4_t*)&sv;
        }
#pragma unroll
        for (int k = 0; k < 4; ++k) a[k] += g[k];
        ((float4_t*)acc)[i] = a;
    }
    if (gid == 0)
        for (long i = nvec << 2; i < n; ++i) {
            float g = (sizeof(SV) == 8)
                ? bf16_to_f32(((const unsigned short*)src_)[i])
                : ((const float*)src_)[i];
            acc[i] += g;
        }
}

struct QU64 { unsigned long long v; };
struct QF128 { float4_t v; };

extern "C" void ps_pack_q8(void* payload, const void* src, long n, int src_dtype,
                           void* stream) {
    long nblk = q8_nblk(n);
    char* q = (char*)payload;
    float* scales = (float*)(q + q8_qbytes(n));
    int blocks; ew_grid(nblk * 64, 256, &blocks);
    if (src_dtype == PS_BF16)
        hipLaunchKernelGGL(pack_q8_kernel<true>, dim3(blocks), dim3(256), 0,
                           (hipStream_t)stream, q, scales, src, n, nblk);
    else
        hipLaunchKernelGGL(pack_q8_kernel<false>, dim3(blocks), dim3(256), 0,
                           (hipStream_t)stream, q, scales, src, n, nblk);
}

extern "C" void ps_unpack_q8(void* dst, const void* payload, long n, int accumulate,
                             void* stream) {
    const char* q = (const char*)payload;
    const float* scales = (const float*)(q + q8_qbytes(n));
    int blocks; ew_grid(n / 4, 256, &blocks);
    if (accumulate)
        hipLaunchKernelGGL(unpack_q8_kernel<true>, dim3(blocks), dim3(256), 0,
                           (hipStream_t)stream, (float*)dst, q, scales, n);
    else
        hipLaunchKernelGGL(unpack_q8_kernel<false>, dim3(blocks), dim3(256), 0,
                           (hipStream_t)stream, (float*)dst, q, scales, n);
}

extern "C" void ps_acc(void* acc, const void* src, long n, int src_dtype, void* stream) {
    int blocks; ew_grid(n / 4, 256, &blocks);
    if (src_dtype == PS_BF16)
        hipLaunchKernelGGL(acc_kernel<QU64>, dim3(blocks), dim3(256), 0,
                           (hipStream_t)stream, (float*)acc, src, n);
    else
        hipLaunchKernelGGL(acc_kernel<QF128>, dim3(blocks), dim3(256), 0,
                           (hipStream_t)stream, (float*)acc, src, n);
}
