// NHWC bf16 pooling kernels: max-pool fwd/bwd (argmax-gather backward, no
// atomics — deterministic) and global average pool fwd/bwd.
//
// Replaces the at::native pooling the reference reaches through
// F.max_pool2d / avg-pool (ref: src/model_ops/lenet.py:31-32 2x2 pools,
// resnet.py:97 avg_pool2d; the ImageNet stem's 3x3-s2 maxpool). Memory-bound
// elementwise-class ops: 16-B channel-vector loads where C allows, guarded
// scalar tails for LeNet's C=20/50.
#include "common.h"

typedef unsigned short ushort8_t __attribute__((ext_vector_type(8)));

// ---- max pool ----
// thread covers (output pixel, channel octet); arg stores the winning
// window slot (r*S+s, < 256) per (pixel, channel) for the backward gather.
__global__ __launch_bounds__(256) void maxpool_fwd_kernel(
    unsigned short* __restrict__ y, unsigned char* __restrict__ arg,
    const unsigned short* __restrict__ x,
    int Nb, int H, int W, int C, int P, int Q,
    int R, int S, int st, int pd,
    unsigned long long mcn, unsigned long long mq, unsigned long long mpq)
{
    EW_IDX
    const int c8n = (C + 7) >> 3;
    const long total = (long)Nb * P * Q * c8n;
    for (long i = gid; i < total; i += stride) {
        // magic-division decode (per-element / and % were ~30 VALU each)
        unsigned pixu = fdiv_u32((unsigned)i, mcn);
        int cb = (int)((unsigned)i - pixu * (unsigned)c8n) * 8;
        long pix = pixu;
        unsigned nu = fdiv_u32(pixu, mpq);
        unsigned rem = pixu - nu * (unsigned)(P * Q);
        unsigned pu = fdiv_u32(rem, mq);
        int q = (int)(rem - pu * (unsigned)Q);
        int p = (int)pu;
        int n = (int)nu;
        const int cw = (cb + 8 <= C) ? 8 : (C - cb);
        float best[8];
        int bidx[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) { best[u] = -3.4e38f; bidx[u] = 0; }
        for (int r = 0; r < R; ++r) {
            int h = p * st - pd + r;
            if (h < 0 || h >= H) continue;
            for (int s = 0; s < S; ++s) {
                int w = q * st - pd + s;
                if (w < 0 || w >= W) continue;
                const unsigned short* px =
                    x + (((long)n * H + h) * W + w) * C + cb;
                if (cw == 8) {
                    ushort8_t v = *(const ushort8_t*)px;
#pragma unroll
                    for (int u = 0; u < 8; ++u) {
                        float f = bf16_to_f32(v[u]);
                        if (f > best[u]) { best[u] = f; bidx[u] = r * S + s; }
                    }
                } else {
                    for (int u = 0; u < cw; ++u) {
                        float f = bf16_to_f32(px[u]);
                        if (f > best[u]) { best[u] = f; bidx[u] = r * S + s; }
                    }
                }
            }
        }
        unsigned short* yp = y + pix * C + cb;
        unsigned char* ap = arg + pix * C + cb;
        if (cw == 8) {
            ushort8_t o;
#pragma unroll
            for (int u = 0; u < 8; ++u) o[u] = f32_to_bf16(best[u]);
            *(ushort8_t*)yp = o;
            unsigned long long a8 = 0;
#pragma unroll
            for (int u = 0; u < 8; ++u)
                a8 |= (unsigned long long)(bidx[u] & 0xff) << (8 * u);
            *(unsigned long long*)ap = a8;
        } else {
            for (int u = 0; u < cw; ++u) {
                yp[u] = f32_to_bf16(best[u]);
                ap[u] = (unsigned char)bidx[u];
            }
        }
    }
}

// backward by gather: each INPUT pixel sums the dy of the <=ceil(R/st)^2
// windows that could have selected it (deterministic, no atomics).
__global__ __launch_bounds__(256) void maxpool_bwd_kernel(
    unsigned short* __restrict__ dx, const unsigned short* __restrict__ dy,
    const unsigned char* __restrict__ arg,
    int Nb, int H, int W, int C, int P, int Q,
    int R, int S, int st, int pd,
    unsigned long long mcn, unsigned long long mw, unsigned long long mhw)
{
    EW_IDX
    const int c8n = (C + 7) >> 3;
    const long total = (long)Nb * H * W * c8n;
    for (long i = gid; i < total; i += stride) {
        unsigned pixu = fdiv_u32((unsigned)i, mcn);
        int cb = (int)((unsigned)i - pixu * (unsigned)c8n) * 8;
        long pix = pixu;
        unsigned nu = fdiv_u32(pixu, mhw);
        unsigned rem = pixu - nu * (unsigned)(H * W);
        unsigned hu = fdiv_u32(rem, mw);
        int w = (int)(rem - hu * (unsigned)W);
        int h = (int)hu;
        int n = (int)nu;
        const int cw = (cb + 8 <= C) ? 8 : (C - cb);
        float acc[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) acc[u] = 0.f;
        int p_lo = (h + pd - R + st) / st; if (p_lo < 0) p_lo = 0;
        int p_hi = (h + pd) / st; if (p_hi >= P) p_hi = P - 1;
        int q_lo = (w + pd - S + st) / st; if (q_lo < 0) q_lo = 0;
        int q_hi = (w + pd) / st; if (q_hi >= Q) q_hi = Q - 1;
        for (int p = p_lo; p <= p_hi; ++p) {
            int r = h - p * st + pd;
            if (r < 0 || r >= R) continue;
            for (int q = q_lo; q <= q_hi; ++q) {
                int s = w - q * st + pd;
                if (s < 0 || s >= S) continue;
                long opix = ((long)n * P + p) * Q + q;
                const unsigned char slot = (unsigned char)(r * S + s);
                const unsigned char* ap = arg + opix * C + cb;
                const unsigned short* gp = dy + opix * C + cb;
                if (cw == 8) {
                    // one 8-B load instead of 8 scalar byte loads (this
                    // kernel was 16x off stream rate on the stem pool)
                    unsigned long long a8 = *(const unsigned long long*)ap;
                    ushort8_t g8 = *(const ushort8_t*)gp;
#pragma unroll
                    for (int u = 0; u < 8; ++u)
                        if (((a8 >> (8 * u)) & 0xff) == slot)
                            acc[u] += bf16_to_f32(g8[u]);
                } else {
                    for (int u = 0; u < cw; ++u)
                        if (ap[u] == slot) acc[u] += bf16_to_f32(gp[u]);
                }
            }
        }
        unsigned short* dp = dx + pix * C + cb;
        if (cw == 8) {
            ushort8_t o;
#pragma unroll
            for (int u = 0; u < 8; ++u) o[u] = f32_to_bf16(acc[u]);
            *(ushort8_t*)dp = o;
        } else {
            for (int u = 0; u < cw; ++u) dp[u] = f32_to_bf16(acc[u]);
        }
    }
}

// ---- global average pool (adaptive_avg_pool2d(x, 1)) ----
__global__ __launch_bounds__(256) void gavg_fwd_kernel(
    unsigned short* __restrict__ y, const unsigned short* __restrict__ x,
    int Nb, int HW, int C)
{
    EW_IDX
    const int c8n = (C + 7) >> 3;
    const long total = (long)Nb * c8n;
    const float inv = 1.f / (float)HW;
    for (long i = gid; i < total; i += stride) {
        int cb = (int)(i % c8n) * 8;
        int n = (int)(i / c8n);
        const int cw = (cb + 8 <= C) ? 8 : (C - cb);
        float acc[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) acc[u] = 0.f;
        const unsigned short* px = x + (long)n * HW * C + cb;
        for (int t = 0; t < HW; ++t, px += C) {
            if (cw == 8) {
                ushort8_t v = *(const ushort8_t*)px;
#pragma unroll
                for (int u = 0; u < 8; ++u) acc[u] += bf16_to_f32(v[u]);
            } else {
                for (int u = 0; u < cw; ++u) acc[u] += bf16_to_f32(px[u]);
            }
        }
        unsigned short* yp = y + (long)n * C + cb;
        for (int u = 0; u < cw; ++u) yp[u] = f32_to_bf16(acc[u] * inv);
    }
}

__global__ __launch_bounds__(256) void gavg_bwd_kernel(
    unsigned short* __restrict__ dx, const unsigned short* __restrict__ dy,
    int Nb, int HW, int C)
{
    EW_IDX
    const int c8n = (C + 7) >> 3;
    const long total = (long)Nb * HW * c8n;
    const float inv = 1.f / (float)HW;
    for (long i = gid; i < total; i += stride) {
        int cb = (int)(i % c8n) * 8;
        long pix = i / c8n;
        int n = (int)(pix / HW);
        const int cw = (cb + 8 <= C) ? 8 : (C - cb);
        const unsigned short* gp = dy + (long)n * C + cb;
        unsigned short* dp = dx + pix * C + cb;
        if (cw == 8) {
            ushort8_t v = *(const ushort8_t*)gp;
            ushort8_t o;
#pragma unroll
            for (int u = 0; u < 8; ++u)
                o[u] = f32_to_bf16(bf16_to_f32(v[u]) * inv);
            *(ushort8_t*)dp = o;
        } else {
            for (int u = 0; u < cw; ++u)
                dp[u] = f32_to_bf16(bf16_to_f32(gp[u]) * inv);
        }
    }
}

extern "C" void ps_maxpool_fwd(
    void* y, void* arg, const void* x, int Nb, int H, int W, int C,
    int P, int Q, int R, int S, int st, int pd, void* strm)
{
    long total = (long)Nb * P * Q * ((C + 7) >> 3);
    int blocks; ew_grid(total, 256, &blocks);
    hipLaunchKernelGGL(maxpool_fwd_kernel, dim3(blocks), dim3(256), 0,
                       (hipStream_t)strm, (unsigned short*)y,
                       (unsigned char*)arg, (const unsigned short*)x,
                       Nb, H, W, C, P, Q, R, S, st, pd,
                       fdiv_magic((C + 7) >> 3), fdiv_magic(Q),
                       fdiv_magic((long)P * Q));
}

extern "C" void ps_maxpool_bwd(
    void* dx, const void* dy, const void* arg, int Nb, int H, int W, int C,
    int P, int Q, int R, int S, int st, int pd, void* strm)
{
    long total = (long)Nb * H * W * ((C + 7) >> 3);
    int blocks; ew_grid(total, 256, &blocks);
    hipLaunchKernelGGL(maxpool_bwd_kernel, dim3(blocks), dim3(256), 0,
                       (hipStream_t)strm, (unsigned short*)dx,
                       (const unsigned short*)dy, (const unsigned char*)arg,
                       Nb, H, W, C, P, Q, R, S, st, pd,
                       fdiv_magic((C + 7) >> 3), fdiv_magic(W),
                       fdiv_magic((long)H * W));
}

extern "C" void ps_gavgpool_fwd(void* y, const void* x, int Nb, int HW,
                                int C, void* strm)
{
    long total = (long)Nb * ((C + 7) >> 3);
    int blocks; ew_grid(total, 256, &blocks);
    hipLaunchKernelGGL(gavg_fwd_kernel, dim3(blocks), dim3(256), 0,
                       (hipStream_t)strm, (unsigned short*)y,
                       (const unsigned short*)x, Nb, HW, C);
}

extern "C" void ps_gavgpool_bwd(void* dx, const void* dy, int Nb, int HW,
                                int C, void* strm)
{
    long total = (long)Nb * HW * ((C + 7) >> 3);
    int blocks; ew_grid(total, 256, &blocks);
    hipLaunchKernelGGL(gavg_bwd_kernel, dim3(blocks), dim3(256), 0,
                       (hipStream_t)strm, (unsigned short*)dx,
                       (const unsigned short*)dy, Nb, HW, C);
}
