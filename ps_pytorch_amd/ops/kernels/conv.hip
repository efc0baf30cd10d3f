// Implicit-GEMM NHWC bf16 convolution (forward / dgrad / wgrad) on MFMA.
//
// North-star kernels (BASELINE.json): the LeNet/ResNet conv forward-backward
// runs on hand-written gfx950 kernels with MFMA + LDS tiling — the
// CDNA4-native replacement for the cuDNN/THNN convs the reference drives
// through nn.Conv2d (ref: src/model_ops/lenet.py:19-33, resnet.py:19-97).
//
// Formulation (NHWC, all tensors bf16, fp32 accumulate):
//   fwd  : out[n,p,q,k] = sum_{r,s,c} in[n, p*st-pad+r, q*st-pad+s, c] * w[k,r,s,c]
//          GEMM  M=N*P*Q (pixels) x N=K, contraction R*S*C.
//   dgrad: dx[n,h,w,c] = sum_{k,r,s} dout[n,(h+pad-r)/st,(w+pad-s)/st,k] * wT[r,s,c,k]
//          GEMM  M=N*H*W x N=C, contraction R*S*K.  The host passes the
//          weight PRE-TRANSPOSED to [R,S,C,K] (one tiny permute per backward)
//          so the B stage is the same contiguous-in-contraction load as fwd —
//          no in-kernel transpose.
//   wgrad: dw[k,r,s,c] = sum_{n,p,q} dout[n,p,q,k] * in[n,p*st-pad+r,...,c]
//          GEMM  M=K x N=R*S*C, contraction = N*P*Q (huge) -> split-K over
//          pixel chunks into f32 partials + deterministic slab reduce (fixed
//          order, no atomics — replicas must stay bit-identical).
//
// Performance structure (guide §5/§6; PMC/disassembly evidence: profiles/):
//   * fwd/dgrad: 4-wave blocks, 128x128 / 128x64 tiles (wave tile 64x64 or
//     64x32 of mfma_f32_16x16x32_bf16), BK=64; T14 pipeline — ONE register
//     staging set, 2 LDS buffers, one barrier per k-step: write tile t+1 to
//     LDS after the barrier, immediately issue loads of t+2, MFMA tile t
//     while the loads fly. Staging addresses are affine-walked (per-row
//     running pointers + wave-uniform step deltas + precomputed tap-validity
//     bitmasks) — a per-step address rebuild made the kernels VALU-bound.
//   * stride is a template parameter; stride-2 dgrad runs by PARITY CLASS
//     (conv_dgrad2_kernel, one launch for all four classes) at full tile
//     density instead of predicating 3/4 of the MFMAs to zero.
//   * AL template: channel counts with full 64-chunk coverage take
//     unconditional 16-B vector loads (the runtime tail guard was ~1/3 of
//     scalar-pipe traffic); LeNet-style ragged shapes take the guarded path.
//   * wgrad: contraction runs over pixels. Operands stage NATURALLY
//     ([pixel][chan] 16-chan subtiles, 32-B row stride, vector writes) and
//     fragments come back through gfx950's hardware transpose read
//     (__builtin_amdgcn_ds_read_tr16_b64_v4bf16). The 3x3 s1 pad1 family
//     uses conv_wgrad_row_kernel: one r-tap per block, all three s-taps
//     served from shared W+2-wide halo rows at shifted (still 8-B-aligned)
//     LDS positions — 3x the MACs per staged byte, R instead of R*S L2
//     re-reads. Pixel->(n,p,q) decode is shifts when the dims are powers
//     of two (every CIFAR shape), runtime div otherwise.
//   * wgrad grids are XCD-swizzled: all blocks covering the SAME pixel
//     chunk land on the SAME XCD (block b runs on XCD b%8) so re-reads of
//     dout/in come from that XCD's L2, not HBM.
#include "common.h"
#include <stdlib.h>

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef unsigned short ushort8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA_BF16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)

union V16 {                       // one 16-byte staging quantum (8 bf16)
    uint4 u4;
    unsigned short us[8];
    bf16x8_t bv;
};

__device__ __forceinline__ V16 zero16() {
    V16 v; v.u4 = make_uint4(0, 0, 0, 0); return v;
}

// 16B load; AL=true (channel counts divisible by 8 — every ResNet/VGG
// shape) is a single unconditional vector load, AL=false takes the guarded
// scalar path (LeNet C=1/20/50). The per-call branch of a runtime check
// showed up as ~1/3 of the kernels' scalar-pipe traffic.
template <bool AL>
__device__ __forceinline__ V16 load16(const unsigned short* src, int cc, int Cn) {
    V16 v;
    if constexpr (AL) {
        v.u4 = *(const uint4*)src;
    } else if (cc + 8 <= Cn) {
        // in-bounds but possibly misaligned (rows of odd-length tensors,
        // e.g. LeNet's K=500/50): pick the widest aligned access. The
        // branches diverge by row parity, but two masked passes still beat
        // the 8-scalar path this used to take for EVERY chunk whenever
        // Cn % 8 != 0.
        size_t a = (size_t)src;
        if ((a & 15) == 0) {
            v.u4 = *(const uint4*)src;
        } else if ((a & 7) == 0) {
            ((unsigned long long*)&v)[0] = ((const unsigned long long*)src)[0];
            ((unsigned long long*)&v)[1] = ((const unsigned long long*)src)[1];
        } else if ((a & 3) == 0) {
#pragma unroll
            for (int u = 0; u < 4; ++u)
                ((unsigned int*)&v)[u] = ((const unsigned int*)src)[u];
        } else {
#pragma unroll
            for (int u = 0; u < 8; ++u) v.us[u] = src[u];
        }
    } else {
        v = zero16();
        for (int u = 0; u < 8; ++u)
            if (cc + u < Cn) v.us[u] = src[u];
    }
    return v;
}

// ------------------------------------------------------------ fwd / dgrad
//
// One templated GEMM core. TM x TN block tile, 256 threads:
//   TN==64 : 4 waves stacked along M (TM=256), wave tile 64x64
//   TN==128: 2x2 wave grid (TM=128), wave tile 64x64
// DGRAD=false: rows are output pixels, cols K, contraction (r,s,c).
// DGRAD=true : rows are input pixels, cols C, contraction (r,s,k),
//              weights given as wT[R,S,C,K].

// TN==64 tiles drop the 72-pad and XOR-swizzle 8-element granules instead
// (same bank spreading, 11% less LDS) — that fits 2 blocks/CU at 256x64 and
// 3 at 128x64, which the padded layout cannot.
template <int TM, int TN, int NBUF = 2>
struct FwdLds {
    static constexpr int PITCH = 64;   // swizzled granules, no pad (all TN)
    unsigned short A[NBUF][TM][PITCH];
    unsigned short B[NBUF][TN][PITCH];
};

// Granule swizzle σ(row, g): position of logical granule g (16 B) within
// row's 128-B span. Must be a permutation of g for each row (writes).
//
// Mode 1 (r1): g ^ (row&7) ^ ((row>>3)&7). PMC showed ~8 pairwise bank
//   collisions per ds_read_b128 lane group (94M conflict cycles on fwd l3):
//   each b128 group mixes two fq half-sets {rows E1, granule g0} and
//   {rows E2, g0^1}, and this σ maps both onto the SAME 8 bank classes.
// Mode 2 (r2 fix): g ^ (((row>>1)&3)<<1). Bank class = (8*(row&1) + σ)
//   mod 16; per group the even-row lanes give σ(E1,g0) = {0,2,4,6} and
//   σ(E2,g0^1) = {5,7,1,3} (disjoint), same for odd rows — a full 16-class
//   bijection per lane group, i.e. conflict-free by the documented
//   (a/4) mod 64 b128 banking. Verified on hardware via
//   SQ_LDS_BANK_CONFLICT; PS_SWZ=1 selects the old layout for A/B.
template <int SWZM>
__device__ __forceinline__ int gswz(int row, int g) {
    if constexpr (SWZM == 2)
        return g ^ (((row >> 1) & 3) << 1);
    return g ^ (row & 7) ^ ((row >> 3) & 7);
}

// SMALL=true: the whole R*S*C contraction fits one 64-chunk (ResNet stem
// 3*3*3=27, LeNet conv1 5*5*1=25) — flatten (r,s,c) into the contraction
// axis via a per-lane gather table and run ONE k-step instead of R*S, so
// the MFMA utilization is RSC/64 of a full tile instead of C/64 per step.
template <int TM, int TN, int STRIDE, bool DGRAD, bool SMALL = false, bool AL = true,
          int NBUF = 2,   // NBUF=1 only for the one-step (RSC<=64) SMALL path
          int SWZM = 2,   // LDS granule swizzle mode (see gswz)
          bool ACCF = false,  // dgrad: epilogue adds `carry` (residual-fork
                              // grad accumulation fused in — kills the
                              // autograd at::add at every block input fork)
          int CPM = 0,    // SMALL padded-channel gather: 4 = C==4 stem
                          // (two 8-B tap halves), 8 = C%8==0 (quantum
                          // within one tap: single 16-B load), 0 = off
          bool GLDS = false,  // stage via global_load_lds (AL && !SMALL):
                              // drops the ds_write pass + staging VGPRs;
                              // invalid taps load from the zero page and
                              // the granule swizzle moves to the SOURCE
                              // address (self-inverse XOR, guide rule 21)
          bool STATS = false> // fwd epilogue also accumulates per-channel
                              // sum/sumsq of the (rounded) outputs into
                              // stats[tile_m][2*Nout] — the following
                              // BatchNorm's stats pass without re-reading
                              // the activation (bn_fwd_foldfin consumes it)
__global__ __launch_bounds__(256) void conv_gemm_kernel(
    const unsigned short* __restrict__ src,  // fwd: in [Nb,H,W,C]; dgrad: dout [Nb,P,Q,K]
    const unsigned short* __restrict__ wgt,  // fwd: w [K,R,S,C]; dgrad: wT [R,S,C,K]
    const unsigned short* __restrict__ bias, // [K] or null (fwd only)
    unsigned short* __restrict__ dst,        // fwd: out [Nb,P,Q,K]; dgrad: dx [Nb,H,W,C]
    const unsigned short* __restrict__ carry,// [dst shape] or null (ACCF)
    const unsigned short* __restrict__ zpage,// 64 zero shorts (GLDS source
                                             // for invalid taps/cols)
    float* __restrict__ stats,               // [tiles_m][2*Nout] or null
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int pad)
{
    __shared__ __attribute__((aligned(16))) FwdLds<TM, TN, NBUF> lds;
    constexpr int SWZ = SWZM;
    constexpr int AR = TM / 32;        // A rows staged per thread
    constexpr int BR = TN / 32;        // B rows staged per thread
    const int Cin = DGRAD ? K : C;     // contraction channel count
    const int Nout = DGRAD ? C : K;    // output column count
    const long M = DGRAD ? (long)Nb * H * W : (long)Nb * P * Q;
    const int tiles_n = (Nout + TN - 1) / TN;
    const long tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const long m0 = tile_m * TM;
    const int n0 = tile_n * TN;
    const int t = threadIdx.x;
    const int trow = t >> 3;           // 0..31
    const int cc8 = (t & 7) * 8;       // this thread's 8-elem contraction chunk
    // GLDS staging geometry: wave w fills rows [w*TM/4, (w+1)*TM/4) of A
    // (and the TN analog of B) with AR (resp. BR) 1-KiB global_load_lds
    // per wave; lane l covers (row = base + rr*8 + (l>>3), position l&7).
    // The LDS image position p of row r holds granule gswz(r, p) — the
    // swizzle is an involution, so the source granule IS gswz(r, l&7).
    const int glane = t & 63, gwid = t >> 6;
    const int ga_row0 = gwid * (TM / 4) + (glane >> 3);
    const int gb_row0 = gwid * (TN / 4) + (glane >> 3);

    // ---- per-row state (computed once) ----
    // !SMALL: running global pointer pA (advanced by a wave-uniform delta
    // each step — no per-step address rebuild, PMC showed the kernel was
    // VALU-bound on it) + a validity bitmask over the R*S taps, shifted
    // right once per (r,s) advance so "valid now" is always bit 0.
    // SMALL: classic origin descriptors for the one-step gather.
    const unsigned short* pA[AR];
    unsigned long long vm[AR];
    long abase[SMALL ? AR : 1]; int ax[SMALL ? AR : 1], ay[SMALL ? AR : 1];
#pragma unroll
    for (int rr = 0; rr < AR; ++rr) {
        const int lrowA = GLDS ? (ga_row0 + rr * 8) : (trow + 32 * rr);
        const int ac8 = GLDS ? gswz<SWZ>(lrowA, t & 7) * 8 : cc8;
        long m = m0 + lrowA;
        pA[rr] = src; vm[rr] = 0;
        if constexpr (SMALL) { ax[rr] = INT_MIN / 2; ay[rr] = 0; abase[rr] = 0; }
        if (m < M) {
            if constexpr (!DGRAD) {
                int n = (int)(m / ((long)P * Q)); int rem = (int)(m % ((long)P * Q));
                int p = rem / Q, q = rem % Q;
                int h0 = p * STRIDE - pad, w0 = q * STRIDE - pad;
                if constexpr (SMALL) {
                    ax[rr] = h0; ay[rr] = w0;
                    abase[rr] = ((long)(n * H + h0) * W + w0) * C;
                } else {
                    pA[rr] = src + ((long)(n * H + h0) * W + w0) * C + ac8;
                    unsigned long long msk = 0;
                    for (int r = 0; r < R; ++r)
                        for (int s = 0; s < S; ++s)
                            if (h0 + r >= 0 && h0 + r < H
                                && w0 + s >= 0 && w0 + s < W)
                                msk |= 1ull << (r * S + s);
                    vm[rr] = msk;
                }
            } else {
                int n = (int)(m / ((long)H * W)); int rem = (int)(m % ((long)H * W));
                int h = rem / W, w = rem % W;
                // stride-1 only (stride-2 dgrad runs conv_dgrad2_kernel)
                int p0 = h + pad, q0 = w + pad;
                pA[rr] = src + (long)n * P * Q * K + ((long)p0 * Q + q0) * K + ac8;
                unsigned long long msk = 0;
                for (int r = 0; r < R; ++r)
                    for (int s = 0; s < S; ++s)
                        if (p0 - r >= 0 && p0 - r < P
                            && q0 - s >= 0 && q0 - s < Q)
                            msk |= 1ull << (r * S + s);
                vm[rr] = msk;
            }
        }
    }

    // wave grid: every wave owns a 64-row sub-tile; columns split when the
    // block is narrower than 4 waves' worth of rows.
    constexpr int WGM = TM / 64;             // waves along M
    constexpr int WGN = 4 / WGM;             // waves along N
    constexpr int NJ = TN / WGN / 16;        // 16-col mfma tiles per wave
    const int lane = t & 63, wid = t >> 6;
    const int wm = (WGN == 1) ? wid : (wid >> 1);
    const int wn = (WGN == 1) ? 0 : (wid & 1);
    const int fr = lane & 15, fq = lane >> 4;

    f32x4_t acc[4][NJ];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < NJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    const int KC = (Cin + 63) >> 6;          // contraction chunks per (r,s)
    const int nsteps = SMALL ? (R * S * C + 63) >> 6 : R * S * KC;
    int lr = 0, ls = 0, lc = 0;              // load-pointer step state


    V16 areg[AR], breg[BR];

    // running B pointers + fixed column-valid flags
    const unsigned short* pB[BR];
    bool colv[BR];
#pragma unroll
    for (int rr = 0; rr < BR; ++rr) {
        const int lrowB = GLDS ? (gb_row0 + rr * 8) : (trow + 32 * rr);
        const int bc8 = GLDS ? gswz<SWZ>(lrowB, t & 7) * 8 : cc8;
        int col = n0 + lrowB;
        colv[rr] = col < Nout;
        if (!colv[rr]) col = 0;
        if constexpr (SMALL)
            pB[rr] = wgt + (long)col * R * S * C + bc8;
        else if constexpr (!DGRAD)
            pB[rr] = wgt + (long)col * R * S * C + bc8;   // (r,s,c0)=(0,0,0)
        else
            pB[rr] = wgt + (long)col * K + bc8;
    }
    const int KCm1 = (KC - 1) * 64;

    // stage loads for contraction step (lr, ls, lc*64) into registers,
    // then advance every pointer by the wave-uniform step delta
    auto load_step = [&]() {
        const int cbase = (lc << 6) + cc8;     // contraction offset (tail guard)
        // SMALL: per-lane flattened (r,s,c) gather table for this chunk's 8
        // contraction elements (e = lc*64 + cc8+u); <=3 chunks, a handful
        // of divides each.
        int tre[SMALL ? 8 : 1], tse[SMALL ? 8 : 1], tce[SMALL ? 8 : 1];
        if constexpr (SMALL && CPM == 0) {
#pragma unroll
            for (int u = 0; u < 8; ++u) {
                int e = (lc << 6) + cc8 + u;
                tre[u] = e / (S * C);
                tse[u] = (e / C) % S;
                tce[u] = e % C;
            }
        }
#pragma unroll
        for (int rr = 0; rr < AR; ++rr) {
            if constexpr (SMALL && CPM == 8) {
                // C%8==0: the 8-elem quantum lies within one (r,s) tap —
                // one guarded 16-B aligned load (LeNet conv2: C 20->24)
                const int RSC = R * S * C;
                const int e0 = (lc << 6) + cc8;
                const int tp = e0 / C;
                const int inner = e0 - tp * C;
                const int tr = tp / S, ts = tp - tr * S;
                V16 v = zero16();
                bool rowv = ax[rr] > INT_MIN / 4;
                int hh = ax[rr] + tr, ww = ay[rr] + ts;
                if (rowv && e0 < RSC && hh >= 0 && hh < H
                    && ww >= 0 && ww < W)
                    v.u4 = *(const uint4*)
                        (src + abase[rr] + ((long)tr * W + ts) * C + inner);
                areg[rr] = v;
            } else if constexpr (SMALL && CPM == 4) {
                // padded-channel stem: the 8-elem quantum is exactly two
                // (r,s) taps of 4 channels each — two aligned 8-B loads
                const int RSC = R * S * 4;
                const int e0 = (lc << 6) + cc8;
                const int t0 = e0 >> 2;
                V16 v = zero16();
                bool rowv = ax[rr] > INT_MIN / 4;
#pragma unroll
                for (int half = 0; half < 2; ++half) {
                    int t = t0 + half;
                    int tr = t / S, ts = t - tr * S;
                    int hh = ax[rr] + tr, ww = ay[rr] + ts;
                    if (rowv && e0 + 4 * half < RSC
                        && hh >= 0 && hh < H && ww >= 0 && ww < W)
                        *(unsigned long long*)&v.us[4 * half] =
                            *(const unsigned long long*)
                                (src + abase[rr] + ((long)tr * W + ts) * 4);
                }
                areg[rr] = v;
            } else if constexpr (SMALL) {
                const int RSC = R * S * C;
                V16 v = zero16();
                bool rowv = ax[rr] > INT_MIN / 4;
#pragma unroll
                for (int u = 0; u < 8; ++u) {
                    int hh = ax[rr] + tre[u], ww = ay[rr] + tse[u];
                    if (rowv && (lc << 6) + cc8 + u < RSC
                        && hh >= 0 && hh < H && ww >= 0 && ww < W)
                        v.us[u] = src[abase[rr]
                                      + ((long)tre[u] * W + tse[u]) * C + tce[u]];
                }
                areg[rr] = v;
            } else {
                areg[rr] = (vm[rr] & 1)
                    ? load16<AL>(pA[rr], cbase, Cin) : zero16();
            }
        }
#pragma unroll
        for (int rr = 0; rr < BR; ++rr) {
            if constexpr (SMALL && CPM == 8) {
                // weight rows are RSC-contiguous and RSC%8==0: in-bounds
                // quanta are full 16-B aligned loads
                const int RSC = R * S * C;
                const int e0 = (lc << 6) + cc8;
                V16 v = zero16();
                if (colv[rr] && e0 < RSC)
                    v.u4 = *(const uint4*)(pB[rr] + (lc << 6));
                breg[rr] = v;
            } else if constexpr (SMALL && CPM == 4) {
                const int RSC = R * S * 4;
                const int e0 = (lc << 6) + cc8;
                V16 v = zero16();
#pragma unroll
                for (int half = 0; half < 2; ++half)
                    if (colv[rr] && e0 + 4 * half < RSC)
                        *(unsigned long long*)&v.us[4 * half] =
                            *(const unsigned long long*)
                                (pB[rr] + (lc << 6) + 4 * half);
                breg[rr] = v;
            } else if constexpr (SMALL)
                breg[rr] = colv[rr]
                    ? load16<AL>(pB[rr] + (lc << 6), (lc << 6) + cc8, R * S * C)
                    : zero16();
            else
                breg[rr] = colv[rr] ? load16<AL>(pB[rr], cbase, Cin) : zero16();
        }
        if constexpr (SMALL) {
            ++lc;
        } else {
            long dA, dB;
            bool rs_adv = (++lc == KC);
            if (rs_adv) {
                lc = 0;
                bool rwrap = (++ls == S);
                if (rwrap) { ls = 0; ++lr; }
                if constexpr (!DGRAD) {
                    dA = (long)(rwrap ? (W - S + 1) : 1) * C - KCm1;
                    dB = (long)C - KCm1;
                } else {
                    dA = (long)(rwrap ? (S - 1 - Q) : -1) * K - KCm1;
                    dB = (long)C * K - KCm1;
                }
            } else { dA = 64; dB = 64; }
#pragma unroll
            for (int rr = 0; rr < AR; ++rr) {
                pA[rr] += dA;
                if (rs_adv) vm[rr] >>= 1;
            }
#pragma unroll
            for (int rr = 0; rr < BR; ++rr) pB[rr] += dB;
        }
    };

    auto write_lds = [&](int buf) {
#pragma unroll
        for (int rr = 0; rr < AR; ++rr) {
            int row = trow + 32 * rr;
            *(uint4*)&lds.A[buf][row][gswz<SWZ>(row, t & 7) * 8] = areg[rr].u4;
        }
#pragma unroll
        for (int rr = 0; rr < BR; ++rr) {
            int row = trow + 32 * rr;
            *(uint4*)&lds.B[buf][row][gswz<SWZ>(row, t & 7) * 8] = breg[rr].u4;
        }
    };

    // GLDS staging: issue the next tile's 1-KiB global_load_lds pieces
    // (lane-linear dest = exactly our [row][granule] image with the
    // swizzle folded into the source granule, see setup) then advance the
    // affine pointers. No registers staged, no ds_write pass; the
    // following __syncthreads drains the DMA (hipcc emits vmcnt(0) there).
    auto glds_step = [&](int buf) {
        if constexpr (GLDS && !SMALL) {
#pragma unroll
            for (int rr = 0; rr < AR; ++rr) {
                const unsigned short* gsrc = (vm[rr] & 1) ? pA[rr] : zpage;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)gsrc,
                    (__attribute__((address_space(3))) void*)
                        &lds.A[buf][gwid * (TM / 4) + rr * 8][0], 16, 0, 0);
            }
#pragma unroll
            for (int rr = 0; rr < BR; ++rr) {
                const unsigned short* gsrc = colv[rr] ? pB[rr] : zpage;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)gsrc,
                    (__attribute__((address_space(3))) void*)
                        &lds.B[buf][gwid * (TN / 4) + rr * 8][0], 16, 0, 0);
            }
            long dA, dB;
            bool rs_adv = (++lc == KC);
            if (rs_adv) {
                lc = 0;
                bool rwrap = (++ls == S);
                if (rwrap) { ls = 0; ++lr; }
                if constexpr (!DGRAD) {
                    dA = (long)(rwrap ? (W - S + 1) : 1) * C - KCm1;
                    dB = (long)C - KCm1;
                } else {
                    dA = (long)(rwrap ? (S - 1 - Q) : -1) * K - KCm1;
                    dB = (long)C * K - KCm1;
                }
            } else { dA = 64; dB = 64; }
#pragma unroll
            for (int rr = 0; rr < AR; ++rr) {
                pA[rr] += dA;
                if (rs_adv) vm[rr] >>= 1;
            }
#pragma unroll
            for (int rr = 0; rr < BR; ++rr) pB[rr] += dB;
        }
    };

    auto mfma_step = [&](int buf) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            bf16x8_t a[4], b[NJ];
#pragma unroll
            for (int mi = 0; mi < 4; ++mi) {
                int row = wm * 64 + mi * 16 + fr;
                a[mi] = *(const bf16x8_t*)
                    &lds.A[buf][row][gswz<SWZ>(row, kk * 4 + fq) * 8];
            }
#pragma unroll
            for (int nj = 0; nj < NJ; ++nj) {
                int row = wn * (NJ * 16) + nj * 16 + fr;
                b[nj] = *(const bf16x8_t*)
                    &lds.B[buf][row][gswz<SWZ>(row, kk * 4 + fq) * 8];
            }
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int nj = 0; nj < NJ; ++nj)
                    acc[mi][nj] = MFMA_BF16(a[mi], b[nj], acc[mi][nj]);
        }
    };

    // ---- pipeline: 2 LDS buffers, one reg set, one barrier per step ----
    if constexpr (GLDS && !SMALL) {
        glds_step(0);
        __syncthreads();
        for (int it = 0; it < nsteps; ++it) {
            if (it + 1 < nsteps)
                glds_step((it + 1) & 1);     // flight covers mfma(it)
            mfma_step(it & 1);
            __syncthreads();
        }
    } else {
    load_step();
    write_lds(0);
    if (nsteps > 1) load_step();
    __syncthreads();
    for (int it = 0; it < nsteps; ++it) {
        if (it + 1 < nsteps) {
            write_lds((it + 1) & 1);
            if (it + 2 < nsteps) load_step();
        }
        mfma_step(it & 1);
        __syncthreads();
    }
    }

    // ---- epilogue ----
    // V2 (NBUF==2 variants): the MFMA C-layout leaves each lane holding
    // column fragments, so the direct write is 32-64 scalar ushort stores
    // per lane — a store-ISSUE-bound tail (guide T21). Stage the f32 tile
    // through the (now idle) LDS buffers, read back row-major and emit one
    // 16-B store per 8 columns; the ACCF carry add becomes a 16-B vector
    // load here instead of scalar gathers.
    if constexpr (NBUF == 2) {
        static_assert(sizeof(lds) >= TM * TN * 4, "f32 staging fits LDS");
        float* sf = reinterpret_cast<float*>(&lds);
#pragma unroll
        for (int nj = 0; nj < NJ; ++nj) {
            int cl = wn * (NJ * 16) + nj * 16 + fr;
            float bv = (!DGRAD && bias && n0 + cl < Nout)
                       ? bf16_to_f32(bias[n0 + cl]) : 0.f;
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
            for (int e = 0; e < 4; ++e)
                sf[(wm * 64 + mi * 16 + fq * 4 + e) * TN + cl]
                    = acc[mi][nj][e] + bv;
        }
        __syncthreads();
        constexpr int SEG = TN / 8;        // 8-col segments per tile row
        constexpr int RPP = 256 / SEG;     // tile rows per pass
        const int segc = (t % SEG) * 8;
        const int rl0 = t / SEG;
        const bool vec_ok = (Nout & 7) == 0;   // 16-B-aligned row segments
        float ssum[STATS ? 8 : 1], ssq[STATS ? 8 : 1];
        if constexpr (STATS)
#pragma unroll
            for (int k2 = 0; k2 < 8; ++k2) { ssum[k2] = 0.f; ssq[k2] = 0.f; }
#pragma unroll
        for (int ps = 0; ps < TM / RPP; ++ps) {
            int rl = ps * RPP + rl0;
            long om = m0 + rl;
            if (om >= M) continue;
            int col = n0 + segc;
            f32x4_t a0 = *(const f32x4_t*)&sf[rl * TN + segc];
            f32x4_t a1 = *(const f32x4_t*)&sf[rl * TN + segc + 4];
            float v[8];
#pragma unroll
            for (int k2 = 0; k2 < 4; ++k2) { v[k2] = a0[k2]; v[4 + k2] = a1[k2]; }
            if (vec_ok && col + 8 <= Nout) {
                V16 o;
                if constexpr (ACCF) {
                    V16 cv = *(const V16*)&carry[om * Nout + col];
#pragma unroll
                    for (int k2 = 0; k2 < 8; ++k2)
                        v[k2] += bf16_to_f32(cv.us[k2]);
                }
#pragma unroll
                for (int k2 = 0; k2 < 8; ++k2) o.us[k2] = f32_to_bf16(v[k2]);
                if constexpr (STATS)
#pragma unroll
                    for (int k2 = 0; k2 < 8; ++k2) {
                        float rv = bf16_to_f32(o.us[k2]);   // rounded value
                        ssum[k2] += rv;
                        ssq[k2] += rv * rv;
                    }
                *(uint4*)&dst[om * Nout + col] = o.u4;
            } else {
                for (int k2 = 0; k2 < 8 && col + k2 < Nout; ++k2) {
                    float vv = v[k2];
                    if constexpr (ACCF)
                        vv += bf16_to_f32(carry[om * Nout + col + k2]);
                    unsigned short ov = f32_to_bf16(vv);
                    if constexpr (STATS) {
                        float rv = bf16_to_f32(ov);
                        ssum[k2] += rv;
                        ssq[k2] += rv * rv;
                    }
                    dst[om * Nout + col + k2] = ov;
                }
            }
        }
        if constexpr (STATS) {
            // reduce the RPP per-thread partials of each channel octet
            // through LDS (reuse sf; deterministic fixed tree) and write
            // this block's [2*Nout] stats row (tile_n slices are disjoint)
            __syncthreads();
            float* sp = sf;                       // [RPP][SEG][8] (+ sq)
#pragma unroll
            for (int k2 = 0; k2 < 8; ++k2) {
                sp[(rl0 * SEG + t % SEG) * 8 + k2] = ssum[k2];
                sp[(RPP * SEG + rl0 * SEG + t % SEG) * 8 + k2] = ssq[k2];
            }
            __syncthreads();
            if (rl0 == 0) {
                for (int rr2 = 1; rr2 < RPP; ++rr2)
#pragma unroll
                    for (int k2 = 0; k2 < 8; ++k2) {
                        ssum[k2] += sp[(rr2 * SEG + t % SEG) * 8 + k2];
                        ssq[k2] += sp[(RPP * SEG + rr2 * SEG + t % SEG) * 8 + k2];
                    }
                float* dst2 = stats + tile_m * 2 * (long)Nout + n0 + segc;
                for (int k2 = 0; k2 < 8 && n0 + segc + k2 < Nout; ++k2) {
                    dst2[k2] = ssum[k2];
                    dst2[Nout + k2] = ssq[k2];
                }
            }
        }
    } else {
#pragma unroll
    for (int nj = 0; nj < NJ; ++nj) {
        int col = n0 + wn * (NJ * 16) + nj * 16 + fr;
        float bv = (!DGRAD && bias && col < Nout) ? bf16_to_f32(bias[col]) : 0.f;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int e = 0; e < 4; ++e) {
            long om = m0 + wm * 64 + mi * 16 + fq * 4 + e;
            if (om < M && col < Nout) {
                float v = acc[mi][nj][e] + bv;
                if constexpr (ACCF)
                    v += bf16_to_f32(carry[om * Nout + col]);
                dst[om * Nout + col] = f32_to_bf16(v);
            }
        }
    }
    }
}

// ------------------------------------------------- dgrad, stride 2
//
// Stride-2 dgrad by parity class: dx pixels (h,w) with h%2==hp, w%2==wp form
// a regular subgrid, and only taps with r == (hp+pad) mod 2 and
// s == (wp+pad) mod 2 contribute — so instead of predicating 3/4 of the MFMA
// work to zero (what a direct stride-2 GEMM does), launch 4 kernels, one per
// class, each contracting only its valid (r,s) subset at full tile density.
// All parity classes ride ONE launch: `ends` = inclusive-exclusive prefix
// of per-class block counts, `codes` = packed (hp<<1|wp) per slot — the
// four separate launches were short-pipeline/launch-overhead bound.
template <int TM, int TN, bool AL = true, int SWZM = 2, bool ACCF = false>
__global__ __launch_bounds__(256) void conv_dgrad2_kernel(
    const unsigned short* __restrict__ dout, // [Nb,P,Q,K]
    const unsigned short* __restrict__ wgt,  // wT [R,S,C,K]
    unsigned short* __restrict__ dx,         // [Nb,H,W,C]
    const unsigned short* __restrict__ carry,// [Nb,H,W,C] or null (ACCF)
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int pad, int4 ends, int4 codes)
{
    int bx = blockIdx.x;
    int cls;
    if      (bx < ends.x) cls = 0;
    else if (bx < ends.y) { cls = 1; bx -= ends.x; }
    else if (bx < ends.z) { cls = 2; bx -= ends.y; }
    else                  { cls = 3; bx -= ends.z; }
    const int code = (cls == 0) ? codes.x : (cls == 1) ? codes.y
                     : (cls == 2) ? codes.z : codes.w;
    const int hp = code >> 1, wp = code & 1;
    const int Hc = (H - hp + 1) >> 1, Wc = (W - wp + 1) >> 1;
    __shared__ __attribute__((aligned(16))) FwdLds<TM, TN> lds;
    constexpr int SWZ = SWZM;
    constexpr int AR = TM / 32;
    constexpr int BR = TN / 32;
    const long M = (long)Nb * Hc * Wc;
    const int tiles_n = (C + TN - 1) / TN;
    const long m0 = (bx / tiles_n) * (long)TM;
    const int n0 = (bx % tiles_n) * TN;
    const int t = threadIdx.x;
    const int trow = t >> 3;
    const int cc8 = (t & 7) * 8;

    const int r0 = (hp + pad) & 1, s0 = (wp + pad) & 1;
    const int nr = (R > r0) ? ((R - r0 + 1) >> 1) : 0;
    const int ns = (S > s0) ? ((S - s0 + 1) >> 1) : 0;

    // affine-walked dout pointers + (ri,si) tap-validity masks (cf. the
    // conv_gemm_kernel staging — same VALU-bound fix)
    const unsigned short* pA[AR];
    unsigned int vm[AR];
#pragma unroll
    for (int rr = 0; rr < AR; ++rr) {
        long m = m0 + trow + 32 * rr;
        pA[rr] = dout; vm[rr] = 0;
        if (m < M) {
            int n = (int)(m / ((long)Hc * Wc)); int rem = (int)(m % ((long)Hc * Wc));
            int h = hp + 2 * (rem / Wc), w = wp + 2 * (rem % Wc);
            int p1 = (h + pad - r0) >> 1, q1 = (w + pad - s0) >> 1;
            pA[rr] = dout + (long)n * P * Q * K + ((long)p1 * Q + q1) * K + cc8;
            unsigned int msk = 0;
            for (int ri = 0; ri < nr; ++ri)
                for (int si = 0; si < ns; ++si)
                    if (p1 - ri >= 0 && p1 - ri < P
                        && q1 - si >= 0 && q1 - si < Q)
                        msk |= 1u << (ri * ns + si);
            vm[rr] = msk;
        }
    }

    constexpr int WGM = TM / 64;
    constexpr int WGN = 4 / WGM;
    constexpr int NJ = TN / WGN / 16;
    const int lane = t & 63, wid = t >> 6;
    const int wm = (WGN == 1) ? wid : (wid >> 1);
    const int wn = (WGN == 1) ? 0 : (wid & 1);
    const int fr = lane & 15, fq = lane >> 4;
    f32x4_t acc[4][NJ];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < NJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    const int KC = (K + 63) >> 6;
    const int KCm1 = (KC - 1) * 64;
    const int nsteps = nr * ns * KC;
    int lsi = 0, lkc = 0;
    V16 areg[AR], breg[BR];

    // running B pointers at (r0, s0, k0=0)
    const unsigned short* pB[BR];
    bool colv[BR];
#pragma unroll
    for (int rr = 0; rr < BR; ++rr) {
        int col = n0 + trow + 32 * rr;
        colv[rr] = col < C;
        pB[rr] = wgt + (((long)(r0 * S + s0) * C) + (colv[rr] ? col : 0))
                 * (long)K + cc8;
    }

    auto load_step = [&]() {
        const int kbase = (lkc << 6) + cc8;
#pragma unroll
        for (int rr = 0; rr < AR; ++rr)
            areg[rr] = (vm[rr] & 1) ? load16<AL>(pA[rr], kbase, K) : zero16();
#pragma unroll
        for (int rr = 0; rr < BR; ++rr)
            breg[rr] = colv[rr] ? load16<AL>(pB[rr], kbase, K) : zero16();
        long dA, dB;
        bool rs_adv = (++lkc == KC);
        if (rs_adv) {
            lkc = 0;
            bool rwrap = (++lsi == ns);
            if (rwrap) lsi = 0;
            // s += 2 -> q -= 1; r += 2 -> p -= 1, q += (ns-1)
            dA = (long)(rwrap ? (ns - 1) * (long)K - (long)Q * K : -(long)K) - KCm1;
            dB = (long)(rwrap ? 2 * S - 2 * (ns - 1) : 2) * C * (long)K - KCm1;
        } else { dA = 64; dB = 64; }
#pragma unroll
        for (int rr = 0; rr < AR; ++rr) {
            pA[rr] += dA;
            if (rs_adv) vm[rr] >>= 1;
        }
#pragma unroll
        for (int rr = 0; rr < BR; ++rr) pB[rr] += dB;
    };

    auto write_lds = [&](int buf) {
#pragma unroll
        for (int rr = 0; rr < AR; ++rr) {
            int row = trow + 32 * rr;
            *(uint4*)&lds.A[buf][row][gswz<SWZ>(row, t & 7) * 8] = areg[rr].u4;
        }
#pragma unroll
        for (int rr = 0; rr < BR; ++rr) {
            int row = trow + 32 * rr;
            *(uint4*)&lds.B[buf][row][gswz<SWZ>(row, t & 7) * 8] = breg[rr].u4;
        }
    };

    auto mfma_step = [&](int buf) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            bf16x8_t a[4], b[NJ];
#pragma unroll
            for (int mi = 0; mi < 4; ++mi) {
                int row = wm * 64 + mi * 16 + fr;
                a[mi] = *(const bf16x8_t*)
                    &lds.A[buf][row][gswz<SWZ>(row, kk * 4 + fq) * 8];
            }
#pragma unroll
            for (int nj = 0; nj < NJ; ++nj) {
                int row = wn * (NJ * 16) + nj * 16 + fr;
                b[nj] = *(const bf16x8_t*)
                    &lds.B[buf][row][gswz<SWZ>(row, kk * 4 + fq) * 8];
            }
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int nj = 0; nj < NJ; ++nj)
                    acc[mi][nj] = MFMA_BF16(a[mi], b[nj], acc[mi][nj]);
        }
    };

    if (nsteps > 0) {
        load_step();
        write_lds(0);
        if (nsteps > 1) load_step();
        __syncthreads();
        for (int it = 0; it < nsteps; ++it) {
            if (it + 1 < nsteps) {
                write_lds((it + 1) & 1);
                if (it + 2 < nsteps) load_step();
            }
            mfma_step(it & 1);
            __syncthreads();
        }
    }

    // V2 epilogue: LDS-staged row-major read-back + 16-B stores (see the
    // conv_gemm_kernel epilogue comment); rows scatter to parity positions
    // but columns stay contiguous, so the segment stores are unchanged.
    {
        static_assert(sizeof(lds) >= TM * TN * 4, "f32 staging fits LDS");
        float* sf = reinterpret_cast<float*>(&lds);
#pragma unroll
        for (int nj = 0; nj < NJ; ++nj) {
            int cl = wn * (NJ * 16) + nj * 16 + fr;
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
            for (int e = 0; e < 4; ++e)
                sf[(wm * 64 + mi * 16 + fq * 4 + e) * TN + cl] = acc[mi][nj][e];
        }
        __syncthreads();
        constexpr int SEG = TN / 8;
        constexpr int RPP = 256 / SEG;
        const int segc = (t % SEG) * 8;
        const int rl0 = t / SEG;
        const bool vec_ok = (C & 7) == 0;
#pragma unroll
        for (int ps = 0; ps < TM / RPP; ++ps) {
            int rl = ps * RPP + rl0;
            long om = m0 + rl;
            if (om >= M) continue;
            int col = n0 + segc;
            int n = (int)(om / ((long)Hc * Wc));
            int rem = (int)(om % ((long)Hc * Wc));
            int h = hp + 2 * (rem / Wc), w = wp + 2 * (rem % Wc);
            long oi = ((long)(n * H + h) * W + w) * C + col;
            f32x4_t a0 = *(const f32x4_t*)&sf[rl * TN + segc];
            f32x4_t a1 = *(const f32x4_t*)&sf[rl * TN + segc + 4];
            float v[8];
#pragma unroll
            for (int k2 = 0; k2 < 4; ++k2) { v[k2] = a0[k2]; v[4 + k2] = a1[k2]; }
            if (vec_ok && col + 8 <= C) {
                if constexpr (ACCF) {
                    V16 cv = *(const V16*)&carry[oi];
#pragma unroll
                    for (int k2 = 0; k2 < 8; ++k2) v[k2] += bf16_to_f32(cv.us[k2]);
                }
                V16 o;
#pragma unroll
                for (int k2 = 0; k2 < 8; ++k2) o.us[k2] = f32_to_bf16(v[k2]);
                *(uint4*)&dx[oi] = o.u4;
            } else {
                for (int k2 = 0; k2 < 8 && col + k2 < C; ++k2) {
                    float vv = v[k2];
                    if constexpr (ACCF) vv += bf16_to_f32(carry[oi + k2]);
                    dx[oi + k2] = f32_to_bf16(vv);
                }
            }
        }
    }
}

// ---------------------------------------------------------------- wgrad

// Natural-layout wgrad LDS: both operands stored [pixel][channel] exactly as
// loaded (vector ds_write_b128 staging — no transpose scatter), reorganized
// into 16-channel subtiles of 32-B row stride so fragments come back through
// gfx950's hardware transpose read `ds_read_b64_tr_b16` (guide T10): a
// 16-lane group reads a [4-pixel][16-chan] block, 4 contiguous bf16 per lane
// at its own 8-B-aligned address, and each lane receives its CHANNEL's 4
// pixel values — i.e. the [chan][pixel] MFMA fragment, transposed in HW.
// Subtile stride is padded +16 elements so the 8 staging writes of one
// 8-lane group land on 8 distinct bank classes.
#define WG_SUB 1040                    // 64*16 + 16 pad (ushort elements)
template <int TK>
struct WgradLds {
    unsigned short A[2][TK / 16][WG_SUB];   // [k-subtile][pixel*16+koff]
    unsigned short B[2][4][WG_SUB];         // [c-subtile][pixel*16+coff]
};

// one hardware transpose read: 4 contiguous bf16 at an 8-B-aligned LDS
// address; the 16-lane group's lanes get their column's 4 values. The
// compiler-modeled builtin (ck_tile idiom) — an inline-asm version with a
// tied-operand waitcnt barrier cost ~150 VALU/step in AGPR<->VGPR copies.
// Pixel-row placement permutation for the [pixel][16-chan] subtiles:
// row stride is 32 B, so rows 8 apart land 256 B apart = the same 64-dword
// bank window, and every tr16 read's 32-lane group spans rows {C+j} and
// {C+8+j} -> a guaranteed 2-way conflict on all 16 pairs (measured: 33% of
// wgrad LDS cycles). pswz(r) = r ^ ((r>>3)&1)<<2 satisfies
// phi(r+8) = phi(r)^4 (mod-8 bank class), so the two 4-row windows of any
// read group land on disjoint bank classes, for the shifted s-tap B rows of
// the row-halo kernel too. Applied identically at write and read setup
// (all precomputed — zero inner-loop cost).
__device__ __forceinline__ int pswz(int r) {
    return r ^ (((r >> 3) & 1) << 2);
}

typedef __bf16 bf16x4_t __attribute__((ext_vector_type(4)));
__device__ __forceinline__ bf16x4_t ds_tr16p(const unsigned short* p) {
    typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 v4;
    auto lp = (__attribute__((address_space(3))) v4*)(p);
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(lp);
}

// One block: TK k x 64 c output tile for ONE (r,s), summing the pixel range
// [sid*chunk, ...) of length `chunk`; f32 partial out[sid][K][R*S*C].
// TK=128 for K>=128 layers (halves the scatter cost per MFMA and the
// dout re-reads); TK=64 otherwise.
// POW2: P*Q and Q are powers of two (shift decode); else runtime div.
template <int STRIDE, bool POW2, int TK, bool AL = true>
__global__ __launch_bounds__(256) void conv_wgrad_kernel(
    const unsigned short* __restrict__ dout,  // [Nb,P,Q,K]
    const unsigned short* __restrict__ in,    // [Nb,H,W,C]
    float* __restrict__ partial,              // [SPLIT][K][R*S*C]
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int pad, int split, int chunk,
    int l2pq, int l2q, int per_xcd,
    unsigned long long mpq, unsigned long long mq)
{
    __shared__ __attribute__((aligned(16))) WgradLds<TK> lds;
    constexpr int KH = TK / 64;        // 64-wide k sub-chunks per tile
    constexpr int MI = TK / 32;        // 16-row mfma tiles per wave (k dim)
    const long M = (long)Nb * P * Q;
    const int RSC = R * S * C;
    const int tiles_k = (K + TK - 1) / TK;
    const int tiles_c = (C + 63) >> 6;
    // XCD swizzle: logical index l groups all tiles of one pixel chunk on
    // one XCD (physical block b runs on XCD b%8).
    long l = (long)(blockIdx.x & 7) * per_xcd + (blockIdx.x >> 3);
    const long nlog = (long)tiles_k * tiles_c * R * S * split;
    if (l >= nlog) return;
    int b = (int)l;
    const int k0 = (b % tiles_k) * TK; b /= tiles_k;
    const int c0 = (b % tiles_c) * 64; b /= tiles_c;
    const int rs = b % (R * S); b /= (R * S);
    const int r = rs / S, s = rs % S;
    const int sid = b;
    const long mbeg = (long)sid * chunk;
    const long mend = (mbeg + chunk < M) ? mbeg + chunk : M;

    const int t = threadIdx.x;
    const int trow = t >> 3;
    const int cc8 = (t & 7) * 8;
    const int lane = t & 63, wid = t >> 6;
    const int wm = wid >> 1, wn = wid & 1;
    const int fr = lane & 15, fq = lane >> 4;
    f32x4_t acc[MI][2];
#pragma unroll
    for (int i = 0; i < MI; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    const int nsteps = (int)((mend - mbeg + 63) >> 6);
    long lm = mbeg;                     // load pointer
    V16 areg[2][KH], breg[2];
    // hoisted dout pointers (advance by 64*K per step)
    const unsigned short* aptr[2];
#pragma unroll
    for (int rr = 0; rr < 2; ++rr)
        aptr[rr] = dout + (mbeg + trow + 32 * rr) * K + k0 + cc8;

    auto load_step = [&]() {
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
            long m = lm + trow + 32 * rr;
            V16 bv = zero16();
#pragma unroll
            for (int h = 0; h < KH; ++h) areg[rr][h] = zero16();
            if (m < mend) {
                int n, p, q;
                if constexpr (POW2) {
                    n = (int)(m >> l2pq);
                    int rem = (int)m & ((1 << l2pq) - 1);
                    p = rem >> l2q; q = rem & ((1 << l2q) - 1);
                } else {
                    n = (int)fdiv_u32((unsigned)m, mpq);
                    unsigned rem = (unsigned)m - (unsigned)n * (unsigned)(P * Q);
                    p = (int)fdiv_u32(rem, mq);
                    q = (int)(rem - (unsigned)p * (unsigned)Q);
                }
#pragma unroll
                for (int h = 0; h < KH; ++h)
                    areg[rr][h] = load16<AL>(aptr[rr] + h * 64, k0 + h * 64 + cc8, K);
                int hh = p * STRIDE - pad + r, ww = q * STRIDE - pad + s;
                if (hh >= 0 && hh < H && ww >= 0 && ww < W)
                    bv = load16<AL>(in + ((long)(n * H + hh) * W + ww) * C + c0 + cc8,
                                c0 + cc8, C);
            }
            breg[rr] = bv;
            aptr[rr] += (long)64 * K;
        }
        lm += 64;
    };

    // ---- natural-store / transpose-read staging (see WgradLds) ----
    // byte offsets within the LDS struct; buffer 1 deltas are constexpr
    constexpr unsigned A1 = (TK / 16) * WG_SUB * 2;     // lds.A[1] - lds.A[0]
    constexpr unsigned BOF = 2 * A1;                    // lds.B[0] - lds.A[0]
    constexpr unsigned BB1 = 4 * WG_SUB * 2;            // lds.B[1] - lds.B[0]
    // staging write pointers: pixel midx's 16B lands in subtile cc8>>4 at
    // column offset cc8&15 (a 16B quantum never straddles a subtile)
    unsigned short* wrA[2][KH];
    unsigned short* wrB[2];
#pragma unroll
    for (int rr = 0; rr < 2; ++rr) {
        int midx = trow + 32 * rr;
#pragma unroll
        for (int h = 0; h < KH; ++h) {
            int kk8 = h * 64 + cc8;
            wrA[rr][h] = &lds.A[0][kk8 >> 4][pswz(midx) * 16 + (kk8 & 15)];
        }
        wrB[rr] = &lds.B[0][cc8 >> 4][pswz(midx) * 16 + (cc8 & 15)];
    }
    // transpose-read offsets: frag (mi,kk,half i) = block rows
    // m = kk*32 + fq*8 + i*4 .. +4, cols = 16-chan subtile; lane fr reads
    // 4 contiguous bf16 of row m + (fr>>2), quad fr&3, and receives its
    // channel's column.
    unsigned roA[2][MI][2];
    unsigned roB[2][2][2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            int m = pswz(kk * 32 + fq * 8 + i * 4 + (fr >> 2));
            int qo = (fr & 3) * 4;
#pragma unroll
            for (int mi = 0; mi < MI; ++mi)
                roA[kk][mi][i] = (unsigned)((char*)&lds.A[0][wm * MI + mi]
                                            [m * 16 + qo] - (char*)&lds);
#pragma unroll
            for (int nj = 0; nj < 2; ++nj)
                roB[kk][nj][i] = (unsigned)((char*)&lds.B[0][wn * 2 + nj]
                                            [m * 16 + qo] - (char*)&lds);
        }

    auto write_lds = [&](int buf) {     // literal buf only
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
#pragma unroll
            for (int h = 0; h < KH; ++h)
                *(uint4*)((char*)wrA[rr][h] + (buf ? A1 : 0)) = areg[rr][h].u4;
            *(uint4*)((char*)wrB[rr] + (buf ? BB1 : 0)) = breg[rr].u4;
        }
    };

    union U64x8 { bf16x4_t h[2]; bf16x8_t v; };
    const char* lb = (const char*)&lds;
    auto mfma_step = [&](int buf) {     // literal buf only
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            U64x8 a[MI], bfr[2];
#pragma unroll
            for (int mi = 0; mi < MI; ++mi) {
                a[mi].h[0] = ds_tr16p((const unsigned short*)
                    (lb + roA[kk][mi][0] + (buf ? A1 : 0)));
                a[mi].h[1] = ds_tr16p((const unsigned short*)
                    (lb + roA[kk][mi][1] + (buf ? A1 : 0)));
            }
#pragma unroll
            for (int nj = 0; nj < 2; ++nj) {
                bfr[nj].h[0] = ds_tr16p((const unsigned short*)
                    (lb + roB[kk][nj][0] + (buf ? BB1 : 0)));
                bfr[nj].h[1] = ds_tr16p((const unsigned short*)
                    (lb + roB[kk][nj][1] + (buf ? BB1 : 0)));
            }
#pragma unroll
            for (int mi = 0; mi < MI; ++mi)
#pragma unroll
                for (int nj = 0; nj < 2; ++nj)
                    acc[mi][nj] = MFMA_BF16(a[mi].v, bfr[nj].v, acc[mi][nj]);
        }
    };

    load_step();
    write_lds(0);
    if (nsteps > 1) load_step();
    __syncthreads();
    for (int it = 0; it < nsteps; it += 2) {
        if (it + 1 < nsteps) {
            write_lds(1);
            if (it + 2 < nsteps) load_step();
        }
        mfma_step(0);
        __syncthreads();
        if (it + 1 >= nsteps) break;
        if (it + 2 < nsteps) {
            write_lds(0);
            if (it + 3 < nsteps) load_step();
        }
        mfma_step(1);
        __syncthreads();
    }

    float* dstp = partial + (long)sid * K * RSC;
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int nj = 0; nj < 2; ++nj)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
        int k = k0 + wm * (MI * 16) + mi * 16 + fq * 4 + e;
        int c = c0 + wn * 32 + nj * 16 + fr;
        if (k < K && c < C)
            dstp[(long)k * RSC + (r * S + s) * C + c] = acc[mi][nj][e];
    }
}

// Row-halo wgrad for the dominant 3x3 stride-1 pad-1 family: one block owns
// (k-tile, c-tile, ONE r) and walks whole output rows. Per 32-pixel step it
// stages the dout rows once plus the matching input rows with a 2-pixel
// halo — all S=3 s-taps then read the SAME staged rows at shifted pixel
// positions (a row shift in the [pixel][chan] subtile image keeps the
// transpose-read 8-B alignment; a channel shift would not). 3x the MACs per
// staged byte of the generic kernel, and dout/in leave L2 R times instead
// of R*S times. Requires: stride 1, R=S=3, pad 1 (so H=P, W=Q and the
// (n*H+h) axis is globally linear in the output row index), Q a power of
// two <= 32, P a power of two.
//
// Contraction axis = 32 consecutive output pixels (exactly one
// mfma_f32_16x16x32 depth): ROWS = 32/Q output rows per step; each lane's
// (row, q) split of its contraction index is a compile-run constant.
template <int TK, bool AL>
__global__ __launch_bounds__(256) void conv_wgrad_row_kernel(
    const unsigned short* __restrict__ dout,  // [Nb,P,Q,K]
    const unsigned short* __restrict__ in,    // [Nb,H,W,C]
    float* __restrict__ partial,              // [SPLIT][K][9*C]
    int Nb, int H, int W, int C, int K, int P, int Q,
    int split, int ipr, int l2q, int l2p, int per_xcd)
{
    // ipr: output rows per split chunk (multiple of ROWS)
    constexpr int PAD = 1;
    const int ROWS = 32 >> l2q;                 // output rows per step
    const int HW2 = W + 2;                      // halo row width
    __shared__ __attribute__((aligned(16))) struct {
        unsigned short A[2][TK / 16][32 * 16 + 16];     // dout [k-sub][m*16+o]
        unsigned short B[2][4][48 * 16 + 16];           // in   [c-sub][hp*16+o]
    } lds;                                               // hp = rib*HW2 + j
    constexpr int MI = TK / 32;
    const int tiles_k = (K + TK - 1) / TK;
    const int tiles_c = (C + 63) >> 6;
    long l = (long)(blockIdx.x & 7) * per_xcd + (blockIdx.x >> 3);
    const long nlog = (long)tiles_k * tiles_c * 3 * split;
    if (l >= nlog) return;
    int b = (int)l;
    const int k0 = (b % tiles_k) * TK; b /= tiles_k;
    const int c0 = (b % tiles_c) * 64; b /= tiles_c;
    const int r = b % 3; b /= 3;
    const int sid = b;
    const long rows_total = (long)Nb * P;
    const long row0 = (long)sid * ipr;
    const long row1 = (row0 + ipr < rows_total) ? row0 + ipr : rows_total;

    const int t = threadIdx.x;
    const int cc8 = (t & 7) * 8;
    const int lane = t & 63, wid = t >> 6;
    const int wm = wid >> 1, wn = wid & 1;
    const int fr = lane & 15, fq = lane >> 4;
    f32x4_t acc[3][MI][2];
#pragma unroll
    for (int s = 0; s < 3; ++s)
#pragma unroll
        for (int i = 0; i < MI; ++i)
#pragma unroll
            for (int j = 0; j < 2; ++j) acc[s][i][j] = {0.f, 0.f, 0.f, 0.f};

    const int nsteps = (int)((row1 - row0 + (1 << (5 - l2q)) - 1)
                             >> (5 - l2q));   // 32 pixels/step

    // ---- running pointers ----
    // dout: linear. in rows: (n*H + h) = rowidx + (r - PAD)  (H == P), so
    // one linear pointer too; validity needs h = p + r - PAD in [0, H).
    const unsigned short* pdout = dout + (row0 << l2q) * K + k0 + cc8;
    const long inrow0 = row0 + r - PAD;          // global in-row of rib 0
    const unsigned short* pin = in + inrow0 * W * C + c0 + cc8;
    long prow = row0;                            // current step's first row

    // per-thread static staging table for B: quantum u covers halo pixel
    // (rib, j) channel-octet cc8 (8 threads per pixel as usual)
    // per-thread static staging assignment: quantum u covers halo pixel
    // (rib, j); indices are compile-time per u so everything stays in
    // registers (a compacted runtime-count loop forced these to scratch)
    const int nbq = ROWS * HW2;                  // halo pixels per step
    int tb_rib[5], tb_j[5];
    bool tb_ex[5];                      // quantum exists (gates the write)
    bool tb_v[5];                       // exists AND w in range (gates load)
    long tb_off[5];                     // static in-row element offset
#pragma unroll
    for (int u = 0; u < 5; ++u) {
        int idx = (t >> 3) + 32 * u;
        tb_ex[u] = idx < nbq;
        int ix = tb_ex[u] ? idx : 0;
        tb_rib[u] = ix / HW2; tb_j[u] = ix % HW2;
        int wpx = tb_j[u] - 1;
        tb_v[u] = tb_ex[u] && wpx >= 0 && wpx < W;
        tb_off[u] = ((long)tb_rib[u] * W + (wpx < 0 ? 0 : wpx)) * C;
    }
    // A staging: m = (t>>3) + 32*u covers 32 m-positions... 32 rows of TK:
    // each thread stages TK/64 quanta per m-row group (like generic wgrad)
    V16 aregs[TK / 64], bregs[5];

    auto load_step = [&]() {
        // A: dout[m][k0 + h*64 + cc8], m = t>>3 (32 rows x 8 thr)
        bool mrow_ok = (prow + ((t >> 3) >> l2q)) < rows_total;
#pragma unroll
        for (int h = 0; h < TK / 64; ++h)
            aregs[h] = mrow_ok
                ? load16<AL>(pdout + (long)(t >> 3) * K + h * 64,
                             k0 + h * 64 + cc8, K)
                : zero16();
        // B: halo rows (w-validity and in-row offsets are precomputed;
        // only the h / image-range part depends on the step)
#pragma unroll
        for (int u = 0; u < 5; ++u) {
            int rib = tb_rib[u];
            int p = (int)((prow + rib) & (P - 1));
            int h = p + r - PAD;
            bool v = tb_v[u] && h >= 0 && h < H
                     && (prow + rib) < rows_total;
            bregs[u] = v ? load16<AL>(pin + tb_off[u], c0 + cc8, C)
                         : zero16();
        }
        pdout += (long)32 * K;
        pin += (long)ROWS * W * C;
        prow += ROWS;
    };

    // LDS write/read offsets (subtile layout as the generic tr kernels)
    constexpr unsigned A1 = sizeof(lds.A[0]);
    constexpr unsigned BB1 = sizeof(lds.B[0]);
    unsigned short* wrA[TK / 64];
#pragma unroll
    for (int h = 0; h < TK / 64; ++h) {
        int kk8 = h * 64 + cc8;
        wrA[h] = &lds.A[0][kk8 >> 4][pswz(t >> 3) * 16 + (kk8 & 15)];
    }
    unsigned short* wrB[5];
#pragma unroll
    for (int u = 0; u < 5; ++u) {
        int hp = pswz(tb_rib[u] * HW2 + tb_j[u]);
        wrB[u] = &lds.B[0][cc8 >> 4][hp * 16 + (cc8 & 15)];
    }

    auto write_lds = [&](int buf) {     // literal buf only
#pragma unroll
        for (int h = 0; h < TK / 64; ++h)
            *(uint4*)((char*)wrA[h] + (buf ? A1 : 0)) = aregs[h].u4;
#pragma unroll
        for (int u = 0; u < 5; ++u)
            if (tb_ex[u])
                *(uint4*)((char*)wrB[u] + (buf ? BB1 : 0)) = bregs[u].u4;
    };

    // read offsets: A-frag lane (fr, fq) element e = fq*8 + i*4 + (fr>>2);
    // B-frag for tap s: halo position = rib(e)*HW2 + q(e) + s
    const char* lb = (const char*)&lds;
    unsigned roA[MI][2];
    unsigned roB[3][2][2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
        int e = fq * 8 + i * 4 + (fr >> 2);
        int qo = (fr & 3) * 4;
#pragma unroll
        for (int mi = 0; mi < MI; ++mi)
            roA[mi][i] = (unsigned)((char*)&lds.A[0][wm * MI + mi]
                                    [pswz(e) * 16 + qo] - lb);
        int rib = e >> l2q, q = e & (Q - 1);
#pragma unroll
        for (int s = 0; s < 3; ++s) {
            int hp = pswz(rib * HW2 + q + s);
#pragma unroll
            for (int nj = 0; nj < 2; ++nj)
                roB[s][nj][i] = (unsigned)((char*)&lds.B[0][wn * 2 + nj]
                                           [hp * 16 + qo] - lb);
        }
    }

    union U64x8 { bf16x4_t h[2]; bf16x8_t v; };
    auto mfma_step = [&](int buf) {     // literal buf only
        U64x8 a[MI];
#pragma unroll
        for (int mi = 0; mi < MI; ++mi) {
            a[mi].h[0] = ds_tr16p((const unsigned short*)
                (lb + roA[mi][0] + (buf ? A1 : 0)));
            a[mi].h[1] = ds_tr16p((const unsigned short*)
                (lb + roA[mi][1] + (buf ? A1 : 0)));
        }
#pragma unroll
        for (int s = 0; s < 3; ++s) {
            U64x8 bf[2];
#pragma unroll
            for (int nj = 0; nj < 2; ++nj) {
                bf[nj].h[0] = ds_tr16p((const unsigned short*)
                    (lb + roB[s][nj][0] + (buf ? BB1 : 0)));
                bf[nj].h[1] = ds_tr16p((const unsigned short*)
                    (lb + roB[s][nj][1] + (buf ? BB1 : 0)));
            }
#pragma unroll
            for (int mi = 0; mi < MI; ++mi)
#pragma unroll
                for (int nj = 0; nj < 2; ++nj)
                    acc[s][mi][nj] = MFMA_BF16(a[mi].v, bf[nj].v,
                                               acc[s][mi][nj]);
        }
    };

    load_step();
    write_lds(0);
    if (nsteps > 1) load_step();
    __syncthreads();
    for (int it = 0; it < nsteps; it += 2) {
        if (it + 1 < nsteps) {
            write_lds(1);
            if (it + 2 < nsteps) load_step();
        }
        mfma_step(0);
        __syncthreads();
        if (it + 1 >= nsteps) break;
        if (it + 2 < nsteps) {
            write_lds(0);
            if (it + 3 < nsteps) load_step();
        }
        mfma_step(1);
        __syncthreads();
    }

    const int RSC = 9 * C;
    float* dstp = partial + (long)sid * K * RSC;
#pragma unroll
    for (int s = 0; s < 3; ++s)
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int nj = 0; nj < 2; ++nj)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
        int k = k0 + wm * (MI * 16) + mi * 16 + fq * 4 + e;
        int c = c0 + wn * 32 + nj * 16 + fr;
        if (k < K && c < C)
            dstp[(long)k * RSC + (r * 3 + s) * C + c] = acc[s][mi][nj][e];
    }
}


// GLDS variant of the row-halo wgrad: both operands stage through
// global_load_lds into LINEAR [row][128 B] images — no staging registers,
// no guarded ds_write pass (the register/write form was issue-bound:
// SQ_ACTIVE_INST_ANY 48%). The tr16 fragment reads move to
// octet-swizzled positions; sigma(row, oct) = oct ^ rho(row) with
// rho(e) = (((e>>1)&1) + ((e>>3)&1)*2) << 1 spreads every {C+j, C+8+j}
// 4-row read window over disjoint 8-dword bank blocks (the involution is
// folded into the SOURCE granule, so DMA stays lane-linear). Invalid
// pixels source the zero page. TK = 64 only.
__device__ __forceinline__ int wrho(int e) {
    return (((e >> 1) & 1) + ((e >> 3) & 1) * 2) << 1;
}

template <bool AL>
__global__ __launch_bounds__(256) void conv_wgrad_row_glds_kernel(
    const unsigned short* __restrict__ dout,  // [Nb,P,Q,K]
    const unsigned short* __restrict__ in,    // [Nb,H,W,C]
    float* __restrict__ partial,              // [SPLIT][K][9*C]
    const unsigned short* __restrict__ zpage,
    int Nb, int H, int W, int C, int K, int P, int Q,
    int split, int ipr, int l2q, int l2p, int per_xcd)
{
    constexpr int TK = 64;
    constexpr int PAD = 1;
    const int ROWS = 32 >> l2q;
    const int HW2 = W + 2;
    // linear images: A 32 pixel-rows, B 64 halo rows (>= worst nbq=40;
    // over-staged rows are zero-page sourced and never read)
    __shared__ __attribute__((aligned(16))) struct {
        unsigned short A[2][32][64];
        unsigned short B[2][64][64];
    } lds;
    constexpr unsigned A1 = sizeof(lds.A[0]);
    constexpr unsigned BB1 = sizeof(lds.B[0]);
    constexpr int MI = TK / 32;
    const int tiles_k = (K + TK - 1) / TK;
    const int tiles_c = (C + 63) >> 6;
    long l = (long)(blockIdx.x & 7) * per_xcd + (blockIdx.x >> 3);
    const long nlog = (long)tiles_k * tiles_c * 3 * split;
    if (l >= nlog) return;
    int b = (int)l;
    const int k0 = (b % tiles_k) * TK; b /= tiles_k;
    const int c0 = (b % tiles_c) * 64; b /= tiles_c;
    const int r = b % 3; b /= 3;
    const int sid = b;
    const long rows_total = (long)Nb * P;
    const long row0 = (long)sid * ipr;
    const long row1 = (row0 + ipr < rows_total) ? row0 + ipr : rows_total;

    const int t = threadIdx.x;
    const int lane = t & 63, wid = t >> 6;
    const int wm = wid >> 1, wn = wid & 1;
    const int fr = lane & 15, fq = lane >> 4;
    f32x4_t acc[3][MI][2];
#pragma unroll
    for (int si = 0; si < 3; ++si)
#pragma unroll
        for (int i = 0; i < MI; ++i)
#pragma unroll
            for (int j = 0; j < 2; ++j) acc[si][i][j] = {0.f, 0.f, 0.f, 0.f};

    const int nsteps = (int)((row1 - row0 + (1 << (5 - l2q)) - 1)
                             >> (5 - l2q));

    // ---- staging assignments (per-lane constants) ----
    // A: wave w stages pixels w*8 + (lane>>3), octet (lane&7) ^ rho(pixel)
    const int apix = wid * 8 + (lane >> 3);
    const int aoct = ((lane & 7) ^ wrho(apix));
    const unsigned short* pdout = dout + (row0 << l2q) * K + k0
                                  + (long)apix * K + aoct * 8;
    // B: wave w stages halo rows {w*16 + j*8 + (lane>>3) : j 0..1}
    const long inrow0 = row0 + r - PAD;
    const unsigned short* pin0 = in + inrow0 * W * C + c0;
    int b_rib[2], b_w[2], b_oct[2];
    bool b_wok[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
        int hp = wid * 16 + j * 8 + (lane >> 3);
        b_oct[j] = ((lane & 7) ^ wrho(hp));
        b_rib[j] = hp / HW2;            // halo row -> (in-row, w-1)
        int wpx = hp - b_rib[j] * HW2 - 1;
        b_wok[j] = wpx >= 0 && wpx < W && hp < ROWS * HW2;
        b_w[j] = wpx < 0 ? 0 : wpx;
    }
    long prow = row0;

    auto glds_stage = [&](int buf) {
        {   // A: one 1-KiB piece per wave
            bool ok = (prow + (apix >> l2q)) < rows_total && apix < 32;
            const unsigned short* gsrc = ok ? pdout : zpage;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gsrc,
                (__attribute__((address_space(3))) void*)
                    &lds.A[buf][wid * 8][0], 16, 0, 0);
        }
#pragma unroll
        for (int j = 0; j < 2; ++j) {   // B: two pieces per wave
            int rib = b_rib[j];
            int p = (int)((prow + rib) & (P - 1));
            int h = p + r - PAD;
            bool ok = b_wok[j] && h >= 0 && h < H
                      && (prow + rib) < rows_total;
            const unsigned short* gsrc = ok
                ? pin0 + ((long)rib * W + b_w[j]) * C + b_oct[j] * 8
                : zpage;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gsrc,
                (__attribute__((address_space(3))) void*)
                    &lds.B[buf][wid * 16 + j * 8][0], 16, 0, 0);
        }
        pdout += (long)32 * K;
        pin0 += (long)ROWS * W * C;
        prow += ROWS;
    };

    // ---- tr16 read offsets (octet-swizzled linear image) ----
    const char* lb0 = (const char*)&lds;
    unsigned roA[MI][2];
    unsigned roB[3][2][2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
        int e = fq * 8 + i * 4 + (fr >> 2);
        int qd = fr & 3;                      // quad within 16-ch subtile
#pragma unroll
        for (int mi = 0; mi < MI; ++mi) {
            int o = (wm * MI + mi) * 2 + (qd >> 1);
            roA[mi][i] = (unsigned)((char*)&lds.A[0][e]
                             [(o ^ wrho(e)) * 8 + (qd & 1) * 4] - lb0);
        }
        int rib = e >> l2q, q = e & (Q - 1);
#pragma unroll
        for (int si = 0; si < 3; ++si) {
            int hp = rib * HW2 + q + si;
#pragma unroll
            for (int nj = 0; nj < 2; ++nj) {
                int o = (wn * 2 + nj) * 2 + (qd >> 1);
                roB[si][nj][i] = (unsigned)((char*)&lds.B[0][hp]
                                 [(o ^ wrho(hp)) * 8 + (qd & 1) * 4] - lb0);
            }
        }
    }

    union U64x8 { bf16x4_t h[2]; bf16x8_t v; };
    auto mfma_step = [&](int buf) {
        U64x8 a[MI];
#pragma unroll
        for (int mi = 0; mi < MI; ++mi) {
            a[mi].h[0] = ds_tr16p((const unsigned short*)
                (lb0 + roA[mi][0] + (buf ? A1 : 0)));
            a[mi].h[1] = ds_tr16p((const unsigned short*)
                (lb0 + roA[mi][1] + (buf ? A1 : 0)));
        }
#pragma unroll
        for (int si = 0; si < 3; ++si) {
            U64x8 bf[2];
#pragma unroll
            for (int nj = 0; nj < 2; ++nj) {
                bf[nj].h[0] = ds_tr16p((const unsigned short*)
                    (lb0 + roB[si][nj][0] + (buf ? BB1 : 0)));
                bf[nj].h[1] = ds_tr16p((const unsigned short*)
                    (lb0 + roB[si][nj][1] + (buf ? BB1 : 0)));
            }
#pragma unroll
            for (int mi = 0; mi < MI; ++mi)
#pragma unroll
                for (int nj = 0; nj < 2; ++nj)
                    acc[si][mi][nj] = MFMA_BF16(a[mi].v, bf[nj].v,
                                                acc[si][mi][nj]);
        }
    };

    glds_stage(0);
    __syncthreads();
    for (int it = 0; it < nsteps; ++it) {
        if (it + 1 < nsteps)
            glds_stage((it + 1) & 1);
        mfma_step(it & 1);
        __syncthreads();
    }

    const int RSC = 9 * C;
    float* dstp = partial + (long)sid * K * RSC;
#pragma unroll
    for (int si = 0; si < 3; ++si)
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int nj = 0; nj < 2; ++nj)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
        int k = k0 + wm * (MI * 16) + mi * 16 + fq * 4 + e;
        int c = c0 + wn * 32 + nj * 16 + fr;
        if (k < K && c < C)
            dstp[(long)k * RSC + (r * 3 + si) * C + c] = acc[si][mi][nj][e];
    }
}

// Small-RSC wgrad (ResNet stem 3x3x3=27, LeNet conv1 5x5x1=25): the whole
// flattened (r,s,c) axis fits one 64-column tile, so one block covers every
// tap in a single pass over its pixel chunk — vs the generic kernel's
// R*S separate 64-c tiles at C/64 utilization each.
template <int STRIDE, bool POW2, bool AL = true, int CPM = 0>
__global__ __launch_bounds__(256) void conv_wgrad_small_kernel(
    const unsigned short* __restrict__ dout,  // [Nb,P,Q,K]
    const unsigned short* __restrict__ in,    // [Nb,H,W,C]
    float* __restrict__ partial,              // [SPLIT][K][R*S*C]
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int pad, int split, int chunk,
    int l2pq, int l2q,
    unsigned long long mpq, unsigned long long mq)
{
    __shared__ __attribute__((aligned(16))) WgradLds<64> lds;
    const long M = (long)Nb * P * Q;
    const int RSC = R * S * C;
    const int tiles_k = (K + 63) >> 6;
    const int nc = (RSC + 63) >> 6;           // flattened-rsc 64-chunks
    int b = blockIdx.x;
    const int k0 = (b % tiles_k) * 64; b /= tiles_k;
    const int e0 = (b % nc) * 64; b /= nc;    // this block's rsc chunk
    const int sid = b;
    const long mbeg = (long)sid * chunk;
    const long mend = (mbeg + chunk < M) ? mbeg + chunk : M;

    const int t = threadIdx.x;
    const int trow = t >> 3;
    const int cc8 = (t & 7) * 8;
    const int lane = t & 63, wid = t >> 6;
    const int wm = wid >> 1, wn = wid & 1;
    const int fr = lane & 15, fq = lane >> 4;
    f32x4_t acc[2][2];
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    // per-lane flattened (r,s,c) gather table (as the SMALL fwd path);
    // toff = the lane's constant element offset from the pixel origin.
    // C4: two aligned 8-B tap loads per quantum instead (padded stem).
    int tre[8], tse[8], toff[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
        int e = e0 + cc8 + u;
        if constexpr (CPM == 8) {
            int tp = e / C;
            tre[u] = tp / S;
            tse[u] = tp - tre[u] * S;
            toff[u] = (tre[u] * W + tse[u]) * C + (e - tp * C);
        } else if constexpr (CPM == 4) {
            int tp = e >> 2;
            tre[u] = tp / S;
            tse[u] = tp - tre[u] * S;
            toff[u] = (tre[u] * W + tse[u]) * 4;
        } else {
            tre[u] = e / (S * C);
            tse[u] = (e / C) % S;
            toff[u] = (tre[u] * W + tse[u]) * C + e % C;
        }
    }

    const int nsteps = (int)((mend - mbeg + 63) >> 6);
    long lm = mbeg;
    V16 areg[2], breg[2];

    auto load_step = [&]() {
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
            long m = lm + trow + 32 * rr;
            V16 av = zero16(), bv = zero16();
            if (m < mend) {
                int n, p, q;
                if constexpr (POW2) {
                    n = (int)(m >> l2pq);
                    int rem = (int)m & ((1 << l2pq) - 1);
                    p = rem >> l2q; q = rem & ((1 << l2q) - 1);
                } else {
                    n = (int)fdiv_u32((unsigned)m, mpq);
                    unsigned rem = (unsigned)m - (unsigned)n * (unsigned)(P * Q);
                    p = (int)fdiv_u32(rem, mq);
                    q = (int)(rem - (unsigned)p * (unsigned)Q);
                }
                av = load16<AL>(dout + m * K + k0 + cc8, k0 + cc8, K);
                int h0 = p * STRIDE - pad, w0 = q * STRIDE - pad;
                const unsigned short* pix =
                    in + (long)n * H * W * C + ((long)h0 * W + w0) * C;
                if constexpr (CPM == 8) {
                    // quantum within one tap: single guarded 16-B load
                    int hh = h0 + tre[0], ww = w0 + tse[0];
                    if (e0 + cc8 < RSC
                        && hh >= 0 && hh < H && ww >= 0 && ww < W)
                        bv.u4 = *(const uint4*)(pix + toff[0]);
                } else if constexpr (CPM == 4) {
#pragma unroll
                    for (int half = 0; half < 2; ++half) {
                        int u = 4 * half;
                        int hh = h0 + tre[u], ww = w0 + tse[u];
                        if (e0 + cc8 + u < RSC
                            && hh >= 0 && hh < H && ww >= 0 && ww < W)
                            *(unsigned long long*)&bv.us[u] =
                                *(const unsigned long long*)(pix + toff[u]);
                    }
                } else {
#pragma unroll
                for (int u = 0; u < 8; ++u) {
                    int hh = h0 + tre[u], ww = w0 + tse[u];
                    if (e0 + cc8 + u < RSC
                        && hh >= 0 && hh < H && ww >= 0 && ww < W)
                        bv.us[u] = pix[toff[u]];
                }
                }
            }
            areg[rr] = av; breg[rr] = bv;
        }
        lm += 64;
    };

    // natural-store / transpose-read staging, as conv_wgrad_kernel (TK=64)
    constexpr unsigned A1 = 4 * WG_SUB * 2;     // lds.A[1] - lds.A[0] (bytes)
    constexpr unsigned BB1 = A1;                // lds.B[1] - lds.B[0]
    unsigned short* wrA[2];
    unsigned short* wrB[2];
#pragma unroll
    for (int rr = 0; rr < 2; ++rr) {
        int midx = pswz(trow + 32 * rr);
        wrA[rr] = &lds.A[0][cc8 >> 4][midx * 16 + (cc8 & 15)];
        wrB[rr] = &lds.B[0][cc8 >> 4][midx * 16 + (cc8 & 15)];
    }
    unsigned roA[2][2][2];
    unsigned roB[2][2][2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            int m = pswz(kk * 32 + fq * 8 + i * 4 + (fr >> 2));
            int qo = (fr & 3) * 4;
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
                roA[kk][mi][i] = (unsigned)((char*)&lds.A[0][wm * 2 + mi]
                                            [m * 16 + qo] - (char*)&lds);
#pragma unroll
            for (int nj = 0; nj < 2; ++nj)
                roB[kk][nj][i] = (unsigned)((char*)&lds.B[0][wn * 2 + nj]
                                            [m * 16 + qo] - (char*)&lds);
        }

    auto write_lds = [&](int buf) {     // literal buf only
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
            *(uint4*)((char*)wrA[rr] + (buf ? A1 : 0)) = areg[rr].u4;
            *(uint4*)((char*)wrB[rr] + (buf ? BB1 : 0)) = breg[rr].u4;
        }
    };

    union U64x8 { bf16x4_t h[2]; bf16x8_t v; };
    const char* lb = (const char*)&lds;
    auto mfma_step = [&](int buf) {     // literal buf only
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            U64x8 a[2], bfr[2];
#pragma unroll
            for (int mi = 0; mi < 2; ++mi) {
                a[mi].h[0] = ds_tr16p((const unsigned short*)
                    (lb + roA[kk][mi][0] + (buf ? A1 : 0)));
                a[mi].h[1] = ds_tr16p((const unsigned short*)
                    (lb + roA[kk][mi][1] + (buf ? A1 : 0)));
            }
#pragma unroll
            for (int nj = 0; nj < 2; ++nj) {
                bfr[nj].h[0] = ds_tr16p((const unsigned short*)
                    (lb + roB[kk][nj][0] + (buf ? BB1 : 0)));
                bfr[nj].h[1] = ds_tr16p((const unsigned short*)
                    (lb + roB[kk][nj][1] + (buf ? BB1 : 0)));
            }
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
#pragma unroll
                for (int nj = 0; nj < 2; ++nj)
                    acc[mi][nj] = MFMA_BF16(a[mi].v, bfr[nj].v, acc[mi][nj]);
        }
    };

    load_step();
    write_lds(0);
    if (nsteps > 1) load_step();
    __syncthreads();
    for (int it = 0; it < nsteps; it += 2) {
        if (it + 1 < nsteps) {
            write_lds(1);
            if (it + 2 < nsteps) load_step();
        }
        mfma_step(0);
        __syncthreads();
        if (it + 1 >= nsteps) break;
        if (it + 2 < nsteps) {
            write_lds(0);
            if (it + 3 < nsteps) load_step();
        }
        mfma_step(1);
        __syncthreads();
    }

    float* dstp = partial + (long)sid * K * RSC;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int nj = 0; nj < 2; ++nj)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
        int k = k0 + wm * 32 + mi * 16 + fq * 4 + e;
        int col = e0 + wn * 32 + nj * 16 + fr;
        if (k < K && col < RSC)
            dstp[(long)k * RSC + col] = acc[mi][nj][e];
    }
}

// Stage-1 slab reduce for large split counts: the final fold kernel's
// parallelism is capped by OUTPUT elements (K*RSC/4 lanes — 36 blocks for
// ResNet-18 l1, 0.14 blocks/CU) while its work is split*n reads, which
// left it ~7x off stream rate at split~170. Each (block.x, group=block.y)
// sums its group's slabs in registers and stores into the group's first
// slab (disjoint read/write sets per group: deterministic, in place).
__global__ __launch_bounds__(256) void reduce_slabs_part_kernel(
    float* __restrict__ partial, long n, int nslab, int per)
{
    EW_IDX
    const int g = blockIdx.y;
    const int s0 = g * per;
    const int s1 = (s0 + per < nslab) ? s0 + per : nslab;
    if (s0 >= nslab) return;
    long nv = n >> 2;
    for (long i = gid; i < nv; i += stride) {
        f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
        for (int sI = s0; sI < s1; ++sI) {
            f32x4_t v = *(const f32x4_t*)(partial + (long)sI * n + i * 4);
#pragma unroll
            for (int k = 0; k < 4; ++k) acc[k] += v[k];
        }
        *(f32x4_t*)(partial + (long)s0 * n + i * 4) = acc;
    }
    if (gid == 0 && g == 0)
        for (long i = nv << 2; i < n; ++i) {
            // scalar tail: fold EVERY slab here (groups would race on the
            // same tail words otherwise); the fold stage reads slab 0 only
            float acc = 0.f;
            for (int sI = 0; sI < nslab; ++sI) acc += partial[(long)sI * n + i];
            partial[i] = acc;
        }
}

// deterministic slab reduce: dw[e] = sum_s partial[s*stride_][e] (fixed
// order); float4 lanes — the scalar version ran ~18x off stream rate.
__global__ __launch_bounds__(256) void reduce_slabs_kernel(
    unsigned short* __restrict__ dw, const float* __restrict__ partial,
    long n, int nslab, int sstride, int tail_ready)
{
    EW_IDX
    long nv = n >> 2;
    for (long i = gid; i < nv; i += stride) {
        f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
        for (int s = 0; s < nslab; ++s) {
            f32x4_t v = *(const f32x4_t*)(partial + (long)s * sstride * n + i * 4);
#pragma unroll
            for (int k = 0; k < 4; ++k) acc[k] += v[k];
        }
        ushort4_t o;
#pragma unroll
        for (int k = 0; k < 4; ++k) o[k] = f32_to_bf16(acc[k]);
        *(ushort4_t*)(dw + i * 4) = o;
    }
    if (gid == 0)
        for (long i = nv << 2; i < n; ++i) {
            if (tail_ready) { dw[i] = f32_to_bf16(partial[i]); continue; }
            float acc = 0.f;
            for (int s = 0; s < nslab; ++s)
                acc += partial[(long)s * sstride * n + i];
            dw[i] = f32_to_bf16(acc);
        }
}

// two-stage launch helper: split > 8 goes through the filled stage-1 pass
static void launch_reduce_slabs(unsigned short* dw, float* partial, long n,
                                int split, hipStream_t strm)
{
    int blocks;
    if (split > 8) {
        const int groups = 8;
        const int per = (split + groups - 1) / groups;
        const int ngroups = (split + per - 1) / per;
        ew_grid(n >> 2, 256, &blocks);
        hipLaunchKernelGGL(reduce_slabs_part_kernel,
                           dim3(blocks, ngroups), dim3(256), 0, strm,
                           partial, n, split, per);
        hipLaunchKernelGGL(reduce_slabs_kernel, dim3(blocks), dim3(256), 0,
                           strm, dw, partial, n, ngroups, per, 1);
        return;
    }
    ew_grid(n >> 2, 256, &blocks);
    hipLaunchKernelGGL(reduce_slabs_kernel, dim3(blocks), dim3(256), 0, strm,
                       dw, partial, n, split, 1, 0);
}

// column sum for bias grad: db[k] = sum_m dout[m][k], two-stage —
// many blocks write per-block partials (the 1-block version serialized the
// whole M-scan on one CU: 17 ms for LeNet), then one block folds them in
// fixed order (deterministic).
__global__ __launch_bounds__(256) void colsum_part_kernel(
    float* __restrict__ partial, const unsigned short* __restrict__ dout,
    long M, int K)
{
    // row-major layout (bn_stats-style): thread t covers channel octet
    // kg = t % G of row r = t / G, reading 16-B vectors — the previous
    // column-per-lane form read K-strided scalars (fully uncoalesced,
    // ~20x off stream rate on LeNet's [524288, 50] bias grad).
    const int G = (K + 7) >> 3;
    const int Rr = 256 / G;               // rows per block iteration (>=1)
    const int kg = threadIdx.x % G;
    const int rib = threadIdx.x / G;
    const int kb = kg * 8;
    const int cw = (kb + 8 <= K) ? 8 : (K - kb);
    const bool active = rib < Rr && Rr > 0;
    __shared__ float red[256][9];         // +1 pad (bank stride)
    float acc[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) acc[u] = 0.f;
    if (active && Rr > 0) {
        const long chunk = 4L * Rr;
        const bool aligned = ((K & 7) == 0);
        for (long base = (long)blockIdx.x * chunk + rib; base < M;
             base += (long)gridDim.x * chunk) {
#pragma unroll
            for (int u4 = 0; u4 < 4; ++u4) {
                long r = base + (long)u4 * Rr;
                if (r >= M) continue;
                const unsigned short* p = dout + r * K + kb;
                if (aligned && cw == 8) {
                    ushort8_t v = *(const ushort8_t*)p;
#pragma unroll
                    for (int u = 0; u < 8; ++u) acc[u] += bf16_to_f32(v[u]);
                } else {
                    for (int u = 0; u < cw; ++u) acc[u] += bf16_to_f32(p[u]);
                }
            }
        }
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) red[threadIdx.x][u] = acc[u];
    __syncthreads();
    // deterministic tree fold over rib (log2(Rr) steps, non-pow2 safe)
    int live = Rr > 0 ? Rr : 1;
    while (live > 1) {
        int half = (live + 1) >> 1;
        if (active && rib < live - half) {
#pragma unroll
            for (int u = 0; u < 8; ++u)
                red[threadIdx.x][u] += red[(rib + half) * G + kg][u];
        }
        __syncthreads();
        live = half;
    }
    if (rib == 0) {
        float* dst = partial + (long)blockIdx.x * K + kb;
        for (int u = 0; u < cw; ++u) dst[u] = red[kg][u];
    }
}

__global__ __launch_bounds__(256) void colsum_fold_kernel(
    unsigned short* __restrict__ db, const float* __restrict__ partial,
    int nblk, int K)
{
    // one block per output channel; threads stride the partial rows and
    // tree-reduce in LDS. (The old 1-2-block serial 512-row scan was a
    // dependent-latency chain — 117 us per VGG bias grad.)
    __shared__ float red[256];
    const int k = blockIdx.x;
    float acc = 0.f;
    for (int b = threadIdx.x; b < nblk; b += 256)
        acc += partial[(long)b * K + k];
    red[threadIdx.x] = acc;
    __syncthreads();
#pragma unroll
    for (int w = 128; w; w >>= 1) {
        if ((int)threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
        __syncthreads();
    }
    if (threadIdx.x == 0) db[k] = f32_to_bf16(red[0]);
}

// ---------------------------------------------------------------- C API

// PS_SWZ=1 selects the round-1 LDS swizzle for on-hardware A/B (gswz).
static inline int swz_mode() {
    static int m = -1;
    if (m < 0) { const char* e = getenv("PS_SWZ"); m = e ? atoi(e) : 2; }
    return m;
}

// PS_GLDS=0 reverts the aligned conv GEMMs to register staging (A/B).
static inline int glds_mode() {
    static int m = -1;
    if (m < 0) { const char* e = getenv("PS_GLDS"); m = e ? atoi(e) : 1; }
    return m;
}

// 16-B zero source for GLDS invalid taps/columns (thread-safe one-time
// device alloc; first use happens in warmup, before any graph capture)
static const unsigned short* zpage_ptr() {
    static const unsigned short* z = [] {
        void* p = nullptr;
        (void)hipMalloc(&p, 128);
        (void)hipMemset(p, 0, 128);
        return (const unsigned short*)p;
    }();
    return z;
}

#define LAUNCH_GEMM(TM, TN, ST, DG, SM) LAUNCH_GEMM_NB(TM, TN, ST, DG, SM, 2)
#define LAUNCH_GEMM_NB(TM, TN, ST, DG, SM, NBV)                               \
    do {                                                                      \
        if (swz_mode() == 1) LAUNCH_GEMM_SW(TM, TN, ST, DG, SM, NBV, 1, false); \
        else                 LAUNCH_GEMM_SW(TM, TN, ST, DG, SM, NBV, 2, false); \
    } while (0)
#define LAUNCH_GEMM_ACC(TM, TN, ST, DG, SM, ACV)                              \
    do {                                                                      \
        if (swz_mode() == 1) LAUNCH_GEMM_SW(TM, TN, ST, DG, SM, 2, 1, ACV);   \
        else                 LAUNCH_GEMM_SW(TM, TN, ST, DG, SM, 2, 2, ACV);   \
    } while (0)
#define LAUNCH_GEMM_CV(TM, TN, ST, SM, NBV) LAUNCH_GEMM_CP(TM, TN, ST, SM, NBV, 4)
#define LAUNCH_GEMM_CP(TM, TN, ST, SM, NBV, CPV)                              \
    do {                                                                      \
        long M_ = (long)Nb * P * Q;                                           \
        long grid = ((M_ + TM - 1) / TM) * ((K + TN - 1) / TN);               \
        hipLaunchKernelGGL((conv_gemm_kernel<TM, TN, ST, false, SM, ALV, NBV, \
                                             2, false, CPV>),               \
            dim3((unsigned)grid), dim3(256), 0, (hipStream_t)strm,            \
            (const unsigned short*)src, (const unsigned short*)wgt,           \
            (const unsigned short*)bias, (unsigned short*)dst,                \
            (const unsigned short*)carry, nullptr, nullptr,                   \
            Nb, H, W, C, K, P, Q, R, S, pad);                                 \
    } while (0)
#define LAUNCH_GEMM_GLS(TM, TN, ST, DG, SM, NBV, SWV, ACV, GLV, STV)          \
    do {                                                                      \
        long M_ = DG ? (long)Nb * H * W : (long)Nb * P * Q;                   \
        int Nout_ = DG ? C : K;                                               \
        long grid = ((M_ + TM - 1) / TM) * ((Nout_ + TN - 1) / TN);           \
        hipLaunchKernelGGL((conv_gemm_kernel<TM, TN, ST, DG, SM, ALV, NBV,    \
                                             SWV, ACV, 0, GLV, STV>),         \
            dim3((unsigned)grid), dim3(256), 0, (hipStream_t)strm,            \
            (const unsigned short*)src, (const unsigned short*)wgt,           \
            (const unsigned short*)bias, (unsigned short*)dst,                \
            (const unsigned short*)carry, zpage_ptr(), (float*)stats,         \
            Nb, H, W, C, K, P, Q, R, S, pad);                                 \
    } while (0)
#define LAUNCH_GEMM_GL(TM, TN, ST, DG, SM, NBV, SWV, ACV, GLV)                \
    do {                                                                      \
        if (!(DG) && stats) LAUNCH_GEMM_GLS(TM, TN, ST, DG, SM, NBV, SWV, ACV, GLV, true); \
        else                LAUNCH_GEMM_GLS(TM, TN, ST, DG, SM, NBV, SWV, ACV, GLV, false); \
    } while (0)
#define LAUNCH_GEMM_SW(TM, TN, ST, DG, SM, NBV, SWV, ACV)                     \
    do {                                                                      \
        if constexpr (ALV && !(SM)) {                                         \
            if (glds_mode()) LAUNCH_GEMM_GL(TM, TN, ST, DG, SM, NBV, SWV, ACV, true); \
            else             LAUNCH_GEMM_GL(TM, TN, ST, DG, SM, NBV, SWV, ACV, false); \
        } else {                                                              \
            LAUNCH_GEMM_GL(TM, TN, ST, DG, SM, NBV, SWV, ACV, false);         \
        }                                                                     \
    } while (0)

// stats != null: the epilogue also writes per-channel sum/sumsq partials
// [ceil(M/128)][2*K] for the following BatchNorm (generic NBUF=2 fwd paths
// only — the python side mirrors the dispatch to know when to allocate).
extern "C" void ps_conv_fwd(
    const void* src, const void* wgt, const void* bias, void* dst,
    void* stats,
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int stride, int pad, void* strm)
{
    const void* carry = nullptr;       // fwd never accumulates
    bool al = (C & 63) == 0;           // full 64-chunk contraction coverage
#define FWD_BODY()                                                            \
    do {                                                                      \
        if (R * S > 1 && C == 4 && R * S * 4 <= 64) {   /* padded stem */     \
            if (K <= 32) {          /* skinny K (LeNet-class): TN=32 tile */  \
                if (stride == 1) LAUNCH_GEMM_CV(128, 32, 1, true, 1);         \
                else             LAUNCH_GEMM_CV(128, 32, 2, true, 1);         \
            } else if (stride == 1) LAUNCH_GEMM_CV(128, 64, 1, true, 1);      \
            else             LAUNCH_GEMM_CV(128, 64, 2, true, 1);             \
        } else if (R * S > 1 && C == 4 && R * S * 4 <= 256) {                 \
            if (K <= 32) {                                                    \
                if (stride == 1) LAUNCH_GEMM_CV(128, 32, 1, true, 2);         \
                else             LAUNCH_GEMM_CV(128, 32, 2, true, 2);         \
            } else if (stride == 1) LAUNCH_GEMM_CV(128, 64, 1, true, 2);      \
            else             LAUNCH_GEMM_CV(128, 64, 2, true, 2);             \
        } else if (R * S > 1 && (C & 7) == 0 && (C & 63)                      \
                   && R * S * C <= 768) {        /* padded/ragged C%8 */      \
            if (K <= 32) {                                                    \
                if (stride == 1) LAUNCH_GEMM_CP(128, 32, 1, true, 2, 8);      \
                else             LAUNCH_GEMM_CP(128, 32, 2, true, 2, 8);      \
            } else if (stride == 1) LAUNCH_GEMM_CP(128, 64, 1, true, 2, 8);   \
            else             LAUNCH_GEMM_CP(128, 64, 2, true, 2, 8);          \
        } else if (R * S > 1 && R * S * C <= 64) {  /* one-step: 1 buffer */  \
            if (stride == 1) LAUNCH_GEMM_NB(128, 64, 1, false, true, 1);      \
            else             LAUNCH_GEMM_NB(128, 64, 2, false, true, 1);      \
        } else if (R * S > 1 && R * S * C <= 192) {                           \
            if (stride == 1) LAUNCH_GEMM(128, 64, 1, false, true);            \
            else             LAUNCH_GEMM(128, 64, 2, false, true);            \
        } else if (R * S == 1 && C <= 64) {     /* 1x1: one k-step */         \
            if (stride == 1) LAUNCH_GEMM_NB(128, 64, 1, false, false, 1);     \
            else             LAUNCH_GEMM_NB(128, 64, 2, false, false, 1);     \
        } else if (K >= 128) {                                                \
            if (stride == 1) LAUNCH_GEMM(128, 128, 1, false, false);          \
            else             LAUNCH_GEMM(128, 128, 2, false, false);          \
        } else if (K <= 32) {   /* skinny outputs: 2x the column util */      \
            if (stride == 1) LAUNCH_GEMM(128, 32, 1, false, false);           \
            else             LAUNCH_GEMM(128, 32, 2, false, false);           \
        } else {                                                              \
            if (stride == 1) LAUNCH_GEMM(128, 64, 1, false, false);           \
            else             LAUNCH_GEMM(128, 64, 2, false, false);           \
        }                                                                     \
    } while (0)
    if (al) { constexpr bool ALV = true; FWD_BODY(); }
    else    { constexpr bool ALV = false; FWD_BODY(); }
#undef FWD_BODY
}

// wgt here is the TRANSPOSED weight wT[R,S,C,K] (host permutes once per
// backward; ~us for the largest ResNet tensor).
// carry != null fuses a same-shaped bf16 tensor into the dx epilogue
// (dx = dgrad + carry): the residual-fork gradient accumulation that
// autograd otherwise runs as a separate elementwise add.
extern "C" void ps_conv_dgrad(
    const void* src, const void* wgt, void* dst, const void* carry,
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int stride, int pad, void* strm)
{
    const void* bias = nullptr;
    const void* stats = nullptr;       // fwd-only epilogue feature
    bool al = (K & 63) == 0;           // contraction runs over K
    if (stride == 1) {
#define DG_BODY(ACV)                                                          \
        do {                                                                  \
            if (C >= 128)     LAUNCH_GEMM_ACC(128, 128, 1, true, false, ACV); \
            else if (C <= 32) LAUNCH_GEMM_ACC(128, 32, 1, true, false, ACV);  \
            else              LAUNCH_GEMM_ACC(128, 64, 1, true, false, ACV);  \
        } while (0)
        if (al) { constexpr bool ALV = true;
                  if (carry) DG_BODY(true); else DG_BODY(false); }
        else    { constexpr bool ALV = false;
                  if (carry) DG_BODY(true); else DG_BODY(false); }
#undef DG_BODY
    } else {
        // 4 parity-class launches (see conv_dgrad2_kernel); classes with no
        // contributing (r,s) — e.g. 3 of 4 for a 1x1 stride-2 conv — are
        // covered by one bulk memset instead of scattered zero stores.
        const unsigned short* dout = (const unsigned short*)src;
        bool any_empty = false;
        for (int hp = 0; hp < 2 && hp < H; ++hp)
        for (int wp = 0; wp < 2 && wp < W; ++wp) {
            int r0 = (hp + pad) & 1, s0 = (wp + pad) & 1;
            if (R <= r0 || S <= s0) any_empty = true;
        }
        if (any_empty) {
            // parity classes with no contributing taps: dx = carry (or 0)
            if (carry)
                (void)hipMemcpyAsync(dst, carry, (long)Nb * H * W * C * 2,
                                     hipMemcpyDeviceToDevice,
                                     (hipStream_t)strm);
            else
                (void)hipMemsetAsync(dst, 0, (long)Nb * H * W * C * 2,
                                     (hipStream_t)strm);
        }
        int TM_ = 128, TN_ = (C >= 128) ? 128 : 64;
        int tiles_n_ = (C + TN_ - 1) / TN_;
        int ends_[4] = {0, 0, 0, 0}, codes_[4] = {0, 0, 0, 0};
        int slot = 0, total = 0;
        for (int hp = 0; hp < 2 && hp < H; ++hp)
        for (int wp = 0; wp < 2 && wp < W; ++wp) {
            int r0 = (hp + pad) & 1, s0 = (wp + pad) & 1;
            if (R <= r0 || S <= s0) continue;
            int Hc = (H - hp + 1) >> 1, Wc = (W - wp + 1) >> 1;
            long M_ = (long)Nb * Hc * Wc;
            total += (int)(((M_ + TM_ - 1) / TM_) * tiles_n_);
            ends_[slot] = total;
            codes_[slot] = (hp << 1) | wp;
            ++slot;
        }
        for (int i = slot; i < 4; ++i) { ends_[i] = total; codes_[i] = 0; }
        if (total == 0) return;
        int4 ends = make_int4(ends_[0], ends_[1], ends_[2], ends_[3]);
        int4 codes = make_int4(codes_[0], codes_[1], codes_[2], codes_[3]);
#define DG2_SW(TM, TN, ALV, SWV, ACV)                                         \
        hipLaunchKernelGGL((conv_dgrad2_kernel<TM, TN, ALV, SWV, ACV>),       \
            dim3((unsigned)total), dim3(256), 0, (hipStream_t)strm, dout,     \
            (const unsigned short*)wgt, (unsigned short*)dst,                 \
            (const unsigned short*)carry,                                     \
            Nb, H, W, C, K, P, Q, R, S, pad, ends, codes)
#define DG2_AC(TM, TN, ALV, SWV) do {                                         \
        if (carry) DG2_SW(TM, TN, ALV, SWV, true);                            \
        else       DG2_SW(TM, TN, ALV, SWV, false); } while (0)
#define DG2(TM, TN, ALV) do {                                                 \
        if (swz_mode() == 1) DG2_AC(TM, TN, ALV, 1);                          \
        else                 DG2_AC(TM, TN, ALV, 2); } while (0)
        if (al) { if (C >= 128) DG2(128, 128, true);
                  else          DG2(128, 64, true); }
        else    { if (C >= 128) DG2(128, 128, false);
                  else          DG2(128, 64, false); }
#undef DG2
    }
}

static inline int ilog2_exact(long v) {
    int l = 0; while ((1L << l) < v) ++l;
    return ((1L << l) == v) ? l : -1;
}

extern "C" void ps_conv_wgrad(
    const void* dout, const void* in, void* partial_f32, void* dw,
    int Nb, int H, int W, int C, int K, int P, int Q,
    int R, int S, int stride, int pad, int split, void* strm)
{
    long M = (long)Nb * P * Q;
    long chunk64 = (M + (long)split * 64 - 1) / ((long)split * 64);
    int chunk = (int)(chunk64 * 64);
    if (R * S > 1 && ((C == 4 && R * S * 4 <= 256)
                      || ((C & 7) == 0 && (C & 63) && R * S * C <= 768)
                      || R * S * C <= 192)) {
        // flattened stem/LeNet path (C==4 = padded stem; C%8==0 = padded
        // or naturally-aligned ragged channels, e.g. LeNet conv2 20->24)
        int tiles_k = (K + 63) / 64;
        int nc_s = (R * S * C + 63) / 64;
        long grid_s = (long)tiles_k * nc_s * split;
        int l2pq_ = ilog2_exact((long)P * Q), l2q_ = ilog2_exact(Q);
        bool pw = l2pq_ >= 0 && l2q_ >= 0;
        unsigned long long mpq_ = fdiv_magic((long)P * Q);
        unsigned long long mq_ = fdiv_magic(Q);
#define WGS(ST, PW, CV)                                                       \
        hipLaunchKernelGGL((conv_wgrad_small_kernel<ST, PW, false, CV>),      \
            dim3((unsigned)grid_s), dim3(256), 0, (hipStream_t)strm,          \
            (const unsigned short*)dout, (const unsigned short*)in,           \
            (float*)partial_f32, Nb, H, W, C, K, P, Q, R, S, pad, split,      \
            chunk, l2pq_, l2q_, mpq_, mq_)
        if (C == 4) {
            if (stride == 1) { if (pw) WGS(1, true, 4); else WGS(1, false, 4); }
            else             { if (pw) WGS(2, true, 4); else WGS(2, false, 4); }
        } else if ((C & 7) == 0 && (C & 63)) {
            if (stride == 1) { if (pw) WGS(1, true, 8); else WGS(1, false, 8); }
            else             { if (pw) WGS(2, true, 8); else WGS(2, false, 8); }
        } else {
            if (stride == 1) { if (pw) WGS(1, true, 0); else WGS(1, false, 0); }
            else             { if (pw) WGS(2, true, 0); else WGS(2, false, 0); }
        }
#undef WGS
        long n_ = (long)K * R * S * C;
        launch_reduce_slabs((unsigned short*)dw, (float*)partial_f32, n_,
                            split, (hipStream_t)strm);
        return;
    }
    // 3x3 s1 pad1 family -> row-halo kernel (see conv_wgrad_row_kernel);
    // PS_WG_ROW_OFF=1 falls back to the generic kernel (A/B)
    {
        static int row_en = -1;
        if (row_en < 0) row_en = getenv("PS_WG_ROW_OFF") ? 0 : 1;
        int l2q_ = ilog2_exact(Q), l2pq_ = ilog2_exact((long)P * Q);
        if (row_en && stride == 1 && R == 3 && S == 3 && pad == 1 && P == H
            && Q == W && l2q_ >= 0 && l2pq_ >= 0 && Q <= 32) {
            static int wg_glds = -1;
            if (wg_glds < 0) {
                const char* e = getenv("PS_WG_GLDS");
                wg_glds = e ? atoi(e) : 0;
            }
            const int TKr = 64;     // TK=128 acc pressure costs a wave/SIMD
            int tkr = (K + TKr - 1) / TKr, tcr = (C + 63) / 64;
            long rows_total = (long)Nb * P;
            int ROWS = 32 >> l2q_;
            long ipr_l = (rows_total + split - 1) / split;
            ipr_l = ((ipr_l + ROWS - 1) / ROWS) * ROWS;
            long nlog = (long)tkr * tcr * 3 * split;
            int pxc = (int)((nlog + 7) / 8);
            long grid_r = (long)pxc * 8;
            bool alr = ((C & 63) == 0) && (K % TKr == 0);
#define WGR(TKV, ALV)                                                         \
            hipLaunchKernelGGL((conv_wgrad_row_kernel<TKV, ALV>),             \
                dim3((unsigned)grid_r), dim3(256), 0, (hipStream_t)strm,      \
                (const unsigned short*)dout, (const unsigned short*)in,       \
                (float*)partial_f32, Nb, H, W, C, K, P, Q,                    \
                split, (int)ipr_l, l2q_, 0, pxc)
#define WGRG(ALV)                                                             \
            hipLaunchKernelGGL((conv_wgrad_row_glds_kernel<ALV>),             \
                dim3((unsigned)grid_r), dim3(256), 0, (hipStream_t)strm,      \
                (const unsigned short*)dout, (const unsigned short*)in,       \
                (float*)partial_f32, zpage_ptr(), Nb, H, W, C, K, P, Q,       \
                split, (int)ipr_l, l2q_, 0, pxc)
            // GLDS form stages unconditional 16-B pieces: aligned
            // channel/K counts only (alr); ragged shapes keep the
            // register-staged kernel
            if (wg_glds && alr) WGRG(true);
            else if (alr)       WGR(64, true);
            else                WGR(64, false);
#undef WGRG
#undef WGR
            long n_ = (long)K * 9 * C;
            launch_reduce_slabs((unsigned short*)dw, (float*)partial_f32, n_,
                                split, (hipStream_t)strm);
            return;
        }
    }
    int TK = (K >= 128) ? 128 : 64;
    int tiles_k = (K + TK - 1) / TK, tiles_c = (C + 63) / 64;
    long nlog = (long)tiles_k * tiles_c * R * S * split;
    int per_xcd = (int)((nlog + 7) / 8);
    long grid = (long)per_xcd * 8;
    int l2pq = ilog2_exact((long)P * Q), l2q = ilog2_exact(Q);
    bool pow2 = l2pq >= 0 && l2q >= 0;
    unsigned long long mpq = fdiv_magic((long)P * Q);
    unsigned long long mq = fdiv_magic(Q);
#define WG_LAUNCH(ST, PW, TKV)                                                \
    hipLaunchKernelGGL((conv_wgrad_kernel<ST, PW, TKV, ALV>),                 \
        dim3((unsigned)grid), dim3(256), 0, (hipStream_t)strm,                \
        (const unsigned short*)dout, (const unsigned short*)in,               \
        (float*)partial_f32, Nb, H, W, C, K, P, Q, R, S, pad, split, chunk,   \
        l2pq, l2q, per_xcd, mpq, mq)
#define WG_TK(ST, PW) do { if (TK == 128) WG_LAUNCH(ST, PW, 128);             \
                           else WG_LAUNCH(ST, PW, 64); } while (0)
#define WG_AL(ST, PW) do { if ((C & 63) == 0 && K % TK == 0) {                \
        constexpr bool ALV = true; WG_TK(ST, PW);                             \
    } else { constexpr bool ALV = false; WG_TK(ST, PW); } } while (0)
    if (stride == 1) { if (pow2) WG_AL(1, true); else WG_AL(1, false); }
    else             { if (pow2) WG_AL(2, true); else WG_AL(2, false); }
#undef WG_AL
#undef WG_TK
#undef WG_LAUNCH
    long n = (long)K * R * S * C;
    launch_reduce_slabs((unsigned short*)dw, (float*)partial_f32, n, split,
                        (hipStream_t)strm);
}

// partial_f32 must hold 512*K floats.
extern "C" void ps_conv_bias_grad(
    void* db, const void* dout, void* partial_f32, long M, int K, void* strm)
{
    int nblk = (int)((M + 3) / 4 < 512 ? (M + 3) / 4 : 512);
    hipLaunchKernelGGL(colsum_part_kernel, dim3(nblk), dim3(256), 0,
                       (hipStream_t)strm, (float*)partial_f32,
                       (const unsigned short*)dout, M, K);
    hipLaunchKernelGGL(colsum_fold_kernel, dim3(K), dim3(256), 0,
                       (hipStream_t)strm, (unsigned short*)db,
                       (const float*)partial_f32, nblk, K);
}

// [K][RSC] -> [RSC][K] bf16 transpose for the dgrad wT operand (torch's
// permute().contiguous() on a channels-last weight ran as a ~45us gather
// kernel; this is a plain 32x32 LDS-tiled transpose at stream rate).
__global__ __launch_bounds__(256) void wt_transpose_kernel(
    const unsigned short* __restrict__ w, unsigned short* __restrict__ wt,
    int K, int RSC)
{
    __shared__ unsigned short tile[32][33];
    int tiles_e = (RSC + 31) >> 5;
    int k0 = (blockIdx.x / tiles_e) << 5;
    int e0 = (blockIdx.x % tiles_e) << 5;
    int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;   // 32 x 8
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        int k = k0 + ty + 8 * i, e = e0 + tx;
        tile[ty + 8 * i][tx] =
            (k < K && e < RSC) ? w[(long)k * RSC + e] : (unsigned short)0;
    }
    __syncthreads();
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        int e = e0 + ty + 8 * i, k = k0 + tx;
        if (e < RSC && k < K)
            wt[(long)e * K + k] = tile[tx][ty + 8 * i];
    }
}

extern "C" void ps_wt_transpose(void* wt, const void* w, int K, int RSC,
                                void* strm)
{
    long grid = (long)((K + 31) / 32) * ((RSC + 31) / 32);
    hipLaunchKernelGGL(wt_transpose_kernel, dim3((unsigned)grid), dim3(256), 0,
                       (hipStream_t)strm, (const unsigned short*)w,
                       (unsigned short*)wt, K, RSC);
}
