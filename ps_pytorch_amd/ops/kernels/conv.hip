// Implicit-GEMM NHWC bf16 convolution (forward / dgrad / wgrad) on MFMA.
//
// North-star kernels (BASELINE.json): the LeNet/ResNet conv forward-backward
// runs on hand-written gfx950 kernels with MFMA + LDS tiling — the
// CDNA4-native replacement for the cuDNN/THNN convs the reference drives
// through nn.Conv2d (ref: src/model_ops/lenet.py:19-33, resnet.py:19-97).
//
// Formulation (NHWC, all tensors bf16, fp32 accumulate):
//   fwd  : out[n,p,q,k] = sum_{r,s,c} in[n, p*st-pad+r, q*st-pad+s, c] * w[k,r,s,c]
//          GEMM  M=N*P*Q (pixels) x N=K, inner = R*S*C.
//          Weight [K][R][S][C] is ALREADY the B^T layout the mfma B-fragment
//          wants (lane j=lane&15 reads its k-slice contiguously), so both
//          operands stage as [row][k] tiles with NO transpose.
//   dgrad: dx[n,h,w,c] = sum_{k,r,s} dout[n,(h+pad-r)/st,(w+pad-s)/st,k] * w[k,r,s,c]
//          GEMM  M=N*H*W x N=C, inner = K per (r,s); B tile needs a
//          transpose stage (w rows are k-major).
//   wgrad: dw[k,r,s,c] = sum_{n,p,q} dout[n,p,q,k] * in[n,p*st-pad+r,...,c]
//          GEMM  M=K x N=R*S*C, inner = N*P*Q (huge) -> split-K over pixel
//          chunks into f32 partials + deterministic slab reduce (fixed
//          order, no atomics — replicas must stay bit-identical).
//
// Tiles: fwd/dgrad 128x64 (BK=64), wgrad 64x64; 4 waves/block;
// mfma_f32_16x16x32_bf16; LDS rows padded 64->72 elems (144 B) to break
// the power-of-2 column-read conflict (guide §6 G4).
#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA_BF16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)

#define LDSP 72   // padded LDS row length (bf16 elems): 144 B, 16B-aligned

union V16 {                       // one 16-byte staging quantum (8 bf16)
    uint4 u4;
    unsigned short us[8];
    bf16x8_t bv;
};

__device__ __forceinline__ V16 zero16() {
    V16 v; v.u4 = make_uint4(0, 0, 0, 0); return v;
}

// ---------------------------------------------------------------- forward

struct ConvLds {
    unsigned short A[128][LDSP];   // [pixel][kg]  (kg = c within (r,s))
    unsigned short B[64][LDSP];    // [out-ch][kg]
};

__global__ __launch_bounds__(256) void conv_fwd_kernel(
    const unsigned short* __restrict__ in,   // [Nb,H,W,C]
    const unsigned short* __restrict__ wgt,  // [K,R,S,C]
    const unsigned short* __restrict__ bias, // [K] or null
    unsigned short* __restrict__ out,        // [Nb,P,Q,K]
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int stride, int pad)
{
    __shared__ __attribute__((aligned(16))) ConvLds lds;
    const int M = Nb * P * Q;
    const int tiles_n = (K + 63) >> 6;
    const int tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const int m0 = tile_m * 128, n0 = tile_n * 64;
    const int t = threadIdx.x;
    const int RSC = R * S * C;

    // per-thread A staging descriptors: rows t/8 + {0,32,64,96}
    long abase[4]; int aph[4], apw[4];
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
        int i = (t >> 3) + 32 * rr;
        int m = m0 + i;
        if (m < M) {
            int n = m / (P * Q), rem = m % (P * Q);
            int p = rem / Q, q = rem % Q;
            aph[rr] = p * stride - pad;
            apw[rr] = q * stride - pad;
            abase[rr] = ((long)(n * H + aph[rr]) * W + apw[rr]) * C;
        } else { aph[rr] = INT_MIN / 2; apw[rr] = 0; abase[rr] = 0; }
    }
    const int cc8 = (t & 7) * 8;   // this thread's 8-elem column chunk

    const int lane = t & 63, wid = t >> 6;
    const int wm = wid >> 1, wn = wid & 1;        // 2x2 wave grid
    const int fr = lane & 15, fq = lane >> 4;
    f32x4_t acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    for (int r = 0; r < R; ++r)
    for (int s = 0; s < S; ++s)
    for (int c0 = 0; c0 < C; c0 += 64) {
        // ---- stage A (input gather): 4 rows x 16B per thread ----
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
            int i = (t >> 3) + 32 * rr;
            int hh = aph[rr] + r, ww = apw[rr] + s;
            bool v = hh >= 0 && hh < H && ww >= 0 && ww < W;
            int cc = c0 + cc8;
            V16 val = zero16();
            if (v) {
                const unsigned short* src = in + abase[rr] + (long)(r * W + s) * C + cc;
                if ((C & 7) == 0 && cc + 8 <= C) val.u4 = *(const uint4*)src;
                else for (int u = 0; u < 8; ++u)
                    if (cc + u < C) val.us[u] = src[u];
            }
            *(uint4*)&lds.A[i][cc8] = val.u4;
        }
        // ---- stage B (weights, no transpose): 2 rows x 16B ----
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
            int j = (t >> 3) + 32 * rr;
            int k = n0 + j;
            int cc = c0 + cc8;
            V16 val = zero16();
            if (k < K) {
                const unsigned short* src = wgt + (long)k * RSC + (r * S + s) * C + cc;
                if ((C & 7) == 0 && cc + 8 <= C) val.u4 = *(const uint4*)src;
                else for (int u = 0; u < 8; ++u)
                    if (cc + u < C) val.us[u] = src[u];
            }
            *(uint4*)&lds.B[j][cc8] = val.u4;
        }
        __syncthreads();
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            bf16x8_t a[4], b[2];
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
                a[mi] = *(const bf16x8_t*)&lds.A[wm * 64 + mi * 16 + fr][kk * 32 + fq * 8];
#pragma unroll
            for (int nj = 0; nj < 2; ++nj)
                b[nj] = *(const bf16x8_t*)&lds.B[wn * 32 + nj * 16 + fr][kk * 32 + fq * 8];
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int nj = 0; nj < 2; ++nj)
                    acc[mi][nj] = MFMA_BF16(a[mi], b[nj], acc[mi][nj]);
        }
        __syncthreads();
    }

    // ---- epilogue: C/D map col=lane&15 (k), row=(lane>>4)*4+e (m) ----
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int nj = 0; nj < 2; ++nj) {
        int ok = n0 + wn * 32 + nj * 16 + fr;
        float bv = (bias && ok < K) ? bf16_to_f32(bias[ok]) : 0.f;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
            int om = m0 + wm * 64 + mi * 16 + fq * 4 + e;
            if (om < M && ok < K)
                out[(long)om * K + ok] = f32_to_bf16(acc[mi][nj][e] + bv);
        }
    }
}

// ---------------------------------------------------------------- dgrad

__global__ __launch_bounds__(256) void conv_dgrad_kernel(
    const unsigned short* __restrict__ dout, // [Nb,P,Q,K]
    const unsigned short* __restrict__ wgt,  // [K,R,S,C]
    unsigned short* __restrict__ dx,         // [Nb,H,W,C]
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int stride, int pad)
{
    __shared__ __attribute__((aligned(16))) ConvLds lds;   // A:[pixel][k] B:[c][k]
    const int M = Nb * H * W;
    const int tiles_n = (C + 63) >> 6;
    const int tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const int m0 = tile_m * 128, n0 = tile_n * 64;
    const int t = threadIdx.x;
    const int RSC = R * S * C;

    // per-thread pixel descriptors (input coords)
    int an[4], ah[4], aw[4];
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
        int m = m0 + (t >> 3) + 32 * rr;
        if (m < M) {
            an[rr] = m / (H * W);
            int rem = m % (H * W);
            ah[rr] = rem / W; aw[rr] = rem % W;
        } else an[rr] = -1;
    }
    const int cc8 = (t & 7) * 8;

    const int lane = t & 63, wid = t >> 6;
    const int wm = wid >> 1, wn = wid & 1;
    const int fr = lane & 15, fq = lane >> 4;
    f32x4_t acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    for (int r = 0; r < R; ++r)
    for (int s = 0; s < S; ++s)
    for (int k0 = 0; k0 < K; k0 += 64) {
        // ---- stage A: dout gather, contiguous in k ----
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
            int i = (t >> 3) + 32 * rr;
            V16 val = zero16();
            if (an[rr] >= 0) {
                int ph = ah[rr] + pad - r, pw = aw[rr] + pad - s;
                if (ph >= 0 && pw >= 0 && ph % stride == 0 && pw % stride == 0) {
                    int p = ph / stride, q = pw / stride;
                    if (p < P && q < Q) {
                        int kk = k0 + cc8;
                        const unsigned short* src =
                            dout + ((long)(an[rr] * P + p) * Q + q) * K + kk;
                        if ((K & 7) == 0 && kk + 8 <= K) val.u4 = *(const uint4*)src;
                        else for (int u = 0; u < 8; ++u)
                            if (kk + u < K) val.us[u] = src[u];
                    }
                }
            }
            *(uint4*)&lds.A[i][cc8] = val.u4;
        }
        // ---- stage B: w[k][r][s][c] -> lds.B[c][k] (transpose) ----
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
            int kidx = (t >> 3) + 32 * rr;      // k within tile
            int k = k0 + kidx;
            int c = n0 + cc8;                   // 8 consecutive c
            V16 val = zero16();
            if (k < K) {
                const unsigned short* src = wgt + (long)k * RSC + (r * S + s) * C + c;
                for (int u = 0; u < 8; ++u)
                    if (c + u < C) val.us[u] = src[u];
            }
#pragma unroll
            for (int u = 0; u < 8; ++u)
                lds.B[cc8 + u][kidx] = val.us[u];
        }
        __syncthreads();
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            bf16x8_t a[4], b[2];
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
                a[mi] = *(const bf16x8_t*)&lds.A[wm * 64 + mi * 16 + fr][kk * 32 + fq * 8];
#pragma unroll
            for (int nj = 0; nj < 2; ++nj)
                b[nj] = *(const bf16x8_t*)&lds.B[wn * 32 + nj * 16 + fr][kk * 32 + fq * 8];
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int nj = 0; nj < 2; ++nj)
                    acc[mi][nj] = MFMA_BF16(a[mi], b[nj], acc[mi][nj]);
        }
        __syncthreads();
    }

#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int nj = 0; nj < 2; ++nj)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
        int om = m0 + wm * 64 + mi * 16 + fq * 4 + e;
        int oc = n0 + wn * 32 + nj * 16 + fr;
        if (om < M && oc < C)
            dx[(long)om * C + oc] = f32_to_bf16(acc[mi][nj][e]);
    }
}

// ---------------------------------------------------------------- wgrad

struct WgradLds {
    unsigned short A[64][LDSP];   // [k][pixel]
    unsigned short B[64][LDSP];   // [c][pixel]
};

// One block: 64 k x 64 rsc output tile for ONE (r,s), summing a pixel range
// [split_id*chunk, ...) of length `chunk`; f32 partial out[split][k][rsc].
__global__ __launch_bounds__(256) void conv_wgrad_kernel(
    const unsigned short* __restrict__ dout,  // [Nb,P,Q,K]
    const unsigned short* __restrict__ in,    // [Nb,H,W,C]
    float* __restrict__ partial,              // [SPLIT][K][R*S*C]
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int stride, int pad, int split, int chunk)
{
    __shared__ __attribute__((aligned(16))) WgradLds lds;
    const int M = Nb * P * Q;
    const int RSC = R * S * C;
    const int tiles_k = (K + 63) >> 6;
    const int tiles_c = (C + 63) >> 6;
    // grid: [tiles_k * R*S * tiles_c * split]
    int b = blockIdx.x;
    const int k0 = (b % tiles_k) * 64; b /= tiles_k;
    const int rs = b % (R * S); b /= (R * S);
    const int r = rs / S, s = rs % S;
    const int c0 = (b % tiles_c) * 64; b /= tiles_c;
    const int sid = b;
    const int mbeg = sid * chunk;
    const int mend = min(mbeg + chunk, M);

    const int t = threadIdx.x;
    const int cc8 = (t & 7) * 8;
    const int lane = t & 63, wid = t >> 6;
    const int wm = wid >> 1, wn = wid & 1;
    const int fr = lane & 15, fq = lane >> 4;
    f32x4_t acc[2][2];
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    for (int mb = mbeg; mb < mend; mb += 64) {
        // each thread stages 2 pixels (m = mb + t/8 + {0,32}) for A and B
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
            int midx = (t >> 3) + 32 * rr;
            int m = mb + midx;
            V16 av = zero16(), bv = zero16();
            if (m < mend) {
                int n = m / (P * Q), rem = m % (P * Q);
                int p = rem / Q, q = rem % Q;
                // A: dout[m][k0+cc8 ..] contiguous in k
                {
                    int kk = k0 + cc8;
                    const unsigned short* src = dout + (long)m * K + kk;
                    if ((K & 7) == 0 && kk + 8 <= K) av.u4 = *(const uint4*)src;
                    else for (int u = 0; u < 8; ++u)
                        if (kk + u < K) av.us[u] = src[u];
                }
                // B: in[n, p*st-pad+r, q*st-pad+s, c0+cc8..] contiguous in c
                {
                    int hh = p * stride - pad + r, ww = q * stride - pad + s;
                    if (hh >= 0 && hh < H && ww >= 0 && ww < W) {
                        int cc = c0 + cc8;
                        const unsigned short* src =
                            in + ((long)(n * H + hh) * W + ww) * C + cc;
                        if ((C & 7) == 0 && cc + 8 <= C) bv.u4 = *(const uint4*)src;
                        else for (int u = 0; u < 8; ++u)
                            if (cc + u < C) bv.us[u] = src[u];
                    }
                }
            }
            // transpose scatter into [k][m] / [c][m]
#pragma unroll
            for (int u = 0; u < 8; ++u) {
                lds.A[cc8 + u][midx] = av.us[u];
                lds.B[cc8 + u][midx] = bv.us[u];
            }
        }
        __syncthreads();
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            bf16x8_t a[2], bfr[2];
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
                a[mi] = *(const bf16x8_t*)&lds.A[wm * 32 + mi * 16 + fr][kk * 32 + fq * 8];
#pragma unroll
            for (int nj = 0; nj < 2; ++nj)
                bfr[nj] = *(const bf16x8_t*)&lds.B[wn * 32 + nj * 16 + fr][kk * 32 + fq * 8];
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
#pragma unroll
                for (int nj = 0; nj < 2; ++nj)
                    acc[mi][nj] = MFMA_BF16(a[mi], bfr[nj], acc[mi][nj]);
        }
        __syncthreads();
    }

    float* dst = partial + (long)sid * K * RSC;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int nj = 0; nj < 2; ++nj)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
        int k = k0 + wm * 32 + mi * 16 + fq * 4 + e;
        int c = c0 + wn * 32 + nj * 16 + fr;
        if (k < K && c < C)
            dst[(long)k * RSC + (r * S + s) * C + c] = acc[mi][nj][e];
    }
}

// deterministic slab reduce: dw[e] = sum_s partial[s][e] (fixed order)
__global__ __launch_bounds__(256) void reduce_slabs_kernel(
    unsigned short* __restrict__ dw, const float* __restrict__ partial,
    long n, int nslab)
{
    EW_IDX
    for (long i = gid; i < n; i += stride) {
        float acc = 0.f;
        for (int s = 0; s < nslab; ++s) acc += partial[(long)s * n + i];
        dw[i] = f32_to_bf16(acc);
    }
}

// column sum for bias grad: db[k] = sum_m dout[m][k] (one block per 64 k,
// two-stage in-block tree over a grid-stride m loop; deterministic)
__global__ __launch_bounds__(256) void colsum_kernel(
    unsigned short* __restrict__ db, const unsigned short* __restrict__ dout,
    long M, int K)
{
    __shared__ float red[256];
    int k = blockIdx.x * 64 + (threadIdx.x & 63);
    int part = threadIdx.x >> 6;          // 4 m-partitions
    float acc = 0.f;
    if (k < K)
        for (long m = part; m < M; m += 4)
            acc += bf16_to_f32(dout[m * K + k]);
    red[threadIdx.x] = acc;
    __syncthreads();
    if (part == 0 && k < K)
        db[k] = f32_to_bf16(red[threadIdx.x] + red[threadIdx.x + 64] +
                            red[threadIdx.x + 128] + red[threadIdx.x + 192]);
}

// ---------------------------------------------------------------- C API

extern "C" void ps_conv_fwd(
    const void* in, const void* wgt, const void* bias, void* out,
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int stride, int pad, void* strm)
{
    long M = (long)Nb * P * Q;
    int grid = (int)((M + 127) / 128) * ((K + 63) / 64);
    hipLaunchKernelGGL(conv_fwd_kernel, dim3(grid), dim3(256), 0, (hipStream_t)strm,
                       (const unsigned short*)in, (const unsigned short*)wgt,
                       (const unsigned short*)bias, (unsigned short*)out,
                       Nb, H, W, C, K, P, Q, R, S, stride, pad);
}

extern "C" void ps_conv_dgrad(
    const void* dout, const void* wgt, void* dx,
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int stride, int pad, void* strm)
{
    long M = (long)Nb * H * W;
    int grid = (int)((M + 127) / 128) * ((C + 63) / 64);
    hipLaunchKernelGGL(conv_dgrad_kernel, dim3(grid), dim3(256), 0, (hipStream_t)strm,
                       (const unsigned short*)dout, (const unsigned short*)wgt,
                       (unsigned short*)dx,
                       Nb, H, W, C, K, P, Q, R, S, stride, pad);
}

extern "C" void ps_conv_wgrad(
    const void* dout, const void* in, void* partial_f32, void* dw,
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int stride, int pad, int split, void* strm)
{
    long M = (long)Nb * P * Q;
    long chunk64 = (M + (long)split * 64 - 1) / ((long)split * 64);
    int chunk = (int)(chunk64 * 64);
    int tiles_k = (K + 63) / 64, tiles_c = (C + 63) / 64;
    int grid = tiles_k * R * S * tiles_c * split;
    hipLaunchKernelGGL(conv_wgrad_kernel, dim3(grid), dim3(256), 0, (hipStream_t)strm,
                       (const unsigned short*)dout, (const unsigned short*)in,
                       (float*)partial_f32,
                       Nb, H, W, C, K, P, Q, R, S, stride, pad, split, chunk);
    long n = (long)K * R * S * C;
    int blocks; ew_grid(n, 256, &blocks);
    hipLaunchKernelGGL(reduce_slabs_kernel, dim3(blocks), dim3(256), 0,
                       (hipStream_t)strm, (unsigned short*)dw,
                       (const float*)partial_f32, n, split);
}

extern "C" void ps_conv_bias_grad(
    void* db, const void* dout, long M, int K, void* strm)
{
    hipLaunchKernelGGL(colsum_kernel, dim3((K + 63) / 64), dim3(256), 0,
                       (hipStream_t)strm, (unsigned short*)db,
                       (const unsigned short*)dout, M, K);
}
