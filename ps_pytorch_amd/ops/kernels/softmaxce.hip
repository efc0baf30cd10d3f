// Fused softmax + cross-entropy (mean reduction) for bf16 logits, NHWC-free
// ([M, C] rows). Replaces the at::native softmax / nll kernels the reference
// reaches through nn.CrossEntropyLoss (ref: src/distributed_worker.py:98,
// nn_ops.py:60) on the training hot path.
//
// Shapes in play: C = 10/100 (CIFAR), 1000 (ImageNet-syn), M = batch
// (<= 8192). The op is tiny and memory-bound; the design goal is one clean
// wave-per-row pass (64-lane shuffle reductions, 16-B vector loads) with a
// deterministic scalar fold — no atomics, bitwise-stable across replicas.
//
//   fwd: lse[m] = max_c x[m,c] + log(sum_c exp(x[m,c] - max));
//        row_ws[m] = lse[m] - x[m, t_m];  loss = (1/M) sum_m row_ws[m]
//   bwd: dx[m,c] = (exp(x[m,c] - lse[m]) - [c == t_m]) * dloss / M
#include "common.h"

typedef unsigned short ushort8_t __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
    for (int m = 32; m; m >>= 1)
        v = fmaxf(v, __shfl_xor(v, m));
    return v;
}

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
    for (int m = 32; m; m >>= 1)
        v += __shfl_xor(v, m);
    return v;
}

// one wave per row; 4 waves (256 threads) per block
__global__ __launch_bounds__(256) void softmax_ce_fwd_kernel(
    float* __restrict__ row_ws,            // [M] per-row loss
    float* __restrict__ lse,               // [M]
    const unsigned short* __restrict__ x,  // [M, C] bf16
    const long* __restrict__ target,       // [M]
    long M, int C)
{
    const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= M) return;
    const int lane = threadIdx.x & 63;
    const unsigned short* xr = x + row * C;

    float mx = -3.4e38f;
    for (int base = lane * 8; base < C; base += 64 * 8) {
        if (base + 8 <= C) {
            ushort8_t v = *(const ushort8_t*)(xr + base);
#pragma unroll
            for (int u = 0; u < 8; ++u) mx = fmaxf(mx, bf16_to_f32(v[u]));
        } else {
            for (int u = base; u < C; ++u) mx = fmaxf(mx, bf16_to_f32(xr[u]));
        }
    }
    mx = wave_max(mx);

    float s = 0.f;
    for (int base = lane * 8; base < C; base += 64 * 8) {
        if (base + 8 <= C) {
            ushort8_t v = *(const ushort8_t*)(xr + base);
#pragma unroll
            for (int u = 0; u < 8; ++u) s += __expf(bf16_to_f32(v[u]) - mx);
        } else {
            for (int u = base; u < C; ++u) s += __expf(bf16_to_f32(xr[u]) - mx);
        }
    }
    s = wave_sum(s);

    if (lane == 0) {
        float l = mx + __logf(s);
        lse[row] = l;
        long t = target[row];
        row_ws[row] = l - bf16_to_f32(xr[t]);
    }
}

// deterministic fold: ONE block, fixed-stride per-thread partials + fixed
// tree (replicas must agree bitwise; a [M] f32 sum is ~us).
__global__ __launch_bounds__(256) void rowloss_fold_kernel(
    float* __restrict__ loss, const float* __restrict__ row_ws, long M)
{
    __shared__ float red[256];
    float acc = 0.f;
    for (long i = threadIdx.x; i < M; i += 256) acc += row_ws[i];
    red[threadIdx.x] = acc;
    __syncthreads();
#pragma unroll
    for (int w = 128; w; w >>= 1) {
        if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
        __syncthreads();
    }
    if (threadIdx.x == 0) loss[0] = red[0] / (float)M;
}

__global__ __launch_bounds__(256) void softmax_ce_bwd_kernel(
    unsigned short* __restrict__ dx,       // [M, C] bf16
    const unsigned short* __restrict__ x,  // [M, C] bf16
    const float* __restrict__ lse,         // [M]
    const long* __restrict__ target,       // [M]
    const float* __restrict__ dloss,       // [1]
    long M, int C)
{
    const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= M) return;
    const int lane = threadIdx.x & 63;
    const float g = dloss[0] / (float)M;
    const float l = lse[row];
    const long t = target[row];
    const unsigned short* xr = x + row * C;
    unsigned short* dr = dx + row * C;
    for (int base = lane * 8; base < C; base += 64 * 8) {
        if (base + 8 <= C) {
            ushort8_t v = *(const ushort8_t*)(xr + base);
            ushort8_t o;
#pragma unroll
            for (int u = 0; u < 8; ++u) {
                float p = __expf(bf16_to_f32(v[u]) - l);
                o[u] = f32_to_bf16((p - (base + u == t ? 1.f : 0.f)) * g);
            }
            *(ushort8_t*)(dr + base) = o;
        } else {
            for (int u = base; u < C; ++u) {
                float p = __expf(bf16_to_f32(xr[u]) - l);
                dr[u] = f32_to_bf16((p - (u == t ? 1.f : 0.f)) * g);
            }
        }
    }
}

extern "C" void ps_softmax_ce_fwd(
    void* loss, void* lse, void* row_ws, const void* logits,
    const void* target, long M, int C, void* strm)
{
    long blocks = (M + 3) / 4;
    hipLaunchKernelGGL(softmax_ce_fwd_kernel, dim3((unsigned)blocks),
                       dim3(256), 0, (hipStream_t)strm,
                       (float*)row_ws, (float*)lse,
                       (const unsigned short*)logits, (const long*)target,
                       M, C);
    hipLaunchKernelGGL(rowloss_fold_kernel, dim3(1), dim3(256), 0,
                       (hipStream_t)strm, (float*)loss,
                       (const float*)row_ws, M);
}

extern "C" void ps_softmax_ce_bwd(
    void* dx, const void* logits, const void* lse, const void* target,
    const void* dloss, long M, int C, void* strm)
{
    long blocks = (M + 3) / 4;
    hipLaunchKernelGGL(softmax_ce_bwd_kernel, dim3((unsigned)blocks),
                       dim3(256), 0, (hipStream_t)strm,
                       (unsigned short*)dx, (const unsigned short*)logits,
                       (const float*)lse, (const long*)target,
                       (const float*)dloss, M, C);
}
