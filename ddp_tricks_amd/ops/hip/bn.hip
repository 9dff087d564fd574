// bn.hip — BatchNorm (2d NHWC and 1d NC) forward/backward with fused ReLU
// epilogue, fp32 statistics over bf16 activations.
//
// Both cases reduce over M rows × C channels where M = N*H*W (2d,
// channels_last) or N (1d).  All passes are bf16x8-vectorized (G13: hipcc
// does not auto-vectorize bf16 — scalar loads are 2-2.5× slower) and
// C % 8 == 0 is required on the HIP path (the torch fallback covers the
// rest).  Reductions are deterministic: per-split partials into a
// [2][C][S] fp32 slab (combine waves read each channel's splits
// coalesced), fixed-order combine — no fp atomics, bit-reproducible.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <mutex>
#include <unordered_map>
#include "common.h"

static inline int bn_splits(long M, int nw) {
    long s = M / (nw * 4);
    if (s < 1) s = 1;
    if (s > 1024) s = 1024;
    return (int)s;
}

// partial sums: slab[0][c][s] = sum, slab[1][c][s] = sumsq over rows s::S
__global__ void k_bn_partial(const bf16* __restrict__ x, long M, int C,
                             int S, float* __restrict__ slab) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* red = reinterpret_cast<float*>(smem);   // [2][nw][C]
    const int cpg = C >> 3;
    const int c8 = threadIdx.x % cpg;
    const int walker = threadIdx.x / cpg;
    const int nw = blockDim.x / cpg;
    const int s = blockIdx.x;
    float sum[8] = {}, sq[8] = {};
    for (long r = (long)s * nw + walker; r < M; r += (long)S * nw) {
        s16x8 v = reinterpret_cast<const s16x8*>(x + r * C)[c8];
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = us2f((unsigned short)v[j]);
            sum[j] += f;
            sq[j] = fmaf(f, f, sq[j]);
        }
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        red[walker * C + c8 * 8 + j] = sum[j];
        red[nw * C + walker * C + c8 * 8 + j] = sq[j];
    }
    __syncthreads();
    for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float ts = 0.f, tq = 0.f;
        for (int w = 0; w < nw; ++w) {
            ts += red[w * C + c];
            tq += red[nw * C + w * C + c];
        }
        slab[(long)c * S + s] = ts;
        slab[(long)C * S + (long)c * S + s] = tq;
    }
}

// combine: wave per channel; lanes sweep the S splits coalesced
__global__ void k_bn_combine(const float* __restrict__ slab, int S, int C,
                             long M, const float* __restrict__ gamma,
                             const float* __restrict__ beta,
                             float* __restrict__ running_mean,
                             float* __restrict__ running_var,
                             float momentum, float eps,
                             float* __restrict__ save_mean,
                             float* __restrict__ save_invstd,
                             float* __restrict__ scale,
                             float* __restrict__ shift) {
    int c = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    if (c >= C) return;
    float sum = 0.f, sq = 0.f;
    for (int s = lane; s < S; s += 64) {
        sum += slab[(long)c * S + s];
        sq += slab[(long)C * S + (long)c * S + s];
    }
    sum = wave_reduce_sum(sum);
    sq = wave_reduce_sum(sq);
    if (lane != 0) return;
    float mean = sum / M;
    float var = fmaxf(sq / M - mean * mean, 0.f);
    float invstd = rsqrtf(var + eps);
    save_mean[c] = mean;
    save_invstd[c] = invstd;
    float g = gamma ? gamma[c] : 1.f;
    float b = beta ? beta[c] : 0.f;
    float sc = g * invstd;
    scale[c] = sc;
    shift[c] = b - mean * sc;
    if (running_mean) {
        float unbiased = M > 1 ? var * M / (M - 1) : var;
        running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
        running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
}

__global__ void k_bn_eval_coeffs(const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 const float* __restrict__ running_mean,
                                 const float* __restrict__ running_var,
                                 float eps, int C,
                                 float* __restrict__ scale,
                                 float* __restrict__ shift) {
    int c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float invstd = rsqrtf(running_var[c] + eps);
    float g = gamma ? gamma[c] : 1.f;
    float b = beta ? beta[c] : 0.f;
    float sc = g * invstd;
    scale[c] = sc;
    shift[c] = b - running_mean[c] * sc;
}

// apply: y = relu?(x*scale[c] + shift[c] (+ residual)) — bf16x8.  The
// optional residual add feeds ResNet skip connections without a separate
// elementwise pass.
__global__ void k_bn_apply_v8(const bf16* __restrict__ x, bf16* __restrict__ y,
                              const float* __restrict__ scale,
                              const float* __restrict__ shift,
                              long total_v, int Cv, bool relu,
                              const bf16* __restrict__ resid,
                              unsigned char* __restrict__ mask) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    // 2-deep unroll: both 16 B loads in flight before either is consumed
    // (single-load iterations leave HBM latency exposed — ~5 of 8 TB/s).
    // 4-deep was TRIED AND REVERTED: +32 staged VGPRs cost occupancy —
    // bn_bwd_dx measured 294.9 -> 310.0 us, apply unchanged.
    for (; i + stride < total_v; i += 2 * stride) {
        long i2 = i + stride;
        s16x8 v0 = reinterpret_cast<const s16x8*>(x)[i];
        s16x8 v1 = reinterpret_cast<const s16x8*>(x)[i2];
        s16x8 r0 = resid ? reinterpret_cast<const s16x8*>(resid)[i] : s16x8{};
        s16x8 r1 = resid ? reinterpret_cast<const s16x8*>(resid)[i2] : s16x8{};
        int cv0 = (i % Cv) * 8, cv1 = (i2 % Cv) * 8;
        s16x8 o0, o1;
        unsigned m0 = 0, m1 = 0;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = fmaf(us2f((unsigned short)v0[j]), scale[cv0 + j],
                           shift[cv0 + j]);
            if (resid) f += us2f((unsigned short)r0[j]);
            if (relu) { if (f > 0.f) m0 |= 1u << j; f = fmaxf(f, 0.f); }
            o0[j] = (short)f2us(f);
            float g = fmaf(us2f((unsigned short)v1[j]), scale[cv1 + j],
                           shift[cv1 + j]);
            if (resid) g += us2f((unsigned short)r1[j]);
            if (relu) { if (g > 0.f) m1 |= 1u << j; g = fmaxf(g, 0.f); }
            o1[j] = (short)f2us(g);
        }
        reinterpret_cast<s16x8*>(y)[i] = o0;
        reinterpret_cast<s16x8*>(y)[i2] = o1;
        if (mask) { mask[i] = (unsigned char)m0; mask[i2] = (unsigned char)m1; }
    }
    for (; i < total_v; i += stride) {
        int cv = (i % Cv) * 8;
        s16x8 v = reinterpret_cast<const s16x8*>(x)[i];
        s16x8 rv = resid ? reinterpret_cast<const s16x8*>(resid)[i] : s16x8{};
        s16x8 o;
        unsigned m = 0;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = fmaf(us2f((unsigned short)v[j]), scale[cv + j], shift[cv + j]);
            if (resid) f += us2f((unsigned short)rv[j]);
            if (relu) {
                if (f > 0.f) m |= 1u << j;
                f = fmaxf(f, 0.f);
            }
            o[j] = (short)f2us(f);
        }
        reinterpret_cast<s16x8*>(y)[i] = o;
        if (mask) mask[i] = (unsigned char)m;   // byte per 8-channel group
    }
}

// backward partials: slab[0][c][s] = sum(dy_eff), slab[1][c][s] = sum(dy_eff*xhat)
__global__ void k_bn_bwd_partial(const bf16* __restrict__ x,
                                 const bf16* __restrict__ dy,
                                 const unsigned char* __restrict__ mask,
                                 const float* __restrict__ save_mean,
                                 const float* __restrict__ save_invstd,
                                 long M, int C, int S, bool relu,
                                 float* __restrict__ slab) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* red = reinterpret_cast<float*>(smem);
    const int cpg = C >> 3;
    const int c8 = threadIdx.x % cpg;
    const int walker = threadIdx.x / cpg;
    const int nw = blockDim.x / cpg;
    const int s = blockIdx.x;
    float mean[8], invstd[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        mean[j] = save_mean[c8 * 8 + j];
        invstd[j] = save_invstd[c8 * 8 + j];
    }
    float sum_dy[8] = {}, sum_dyx[8] = {};
    const int cpgv = cpg;
    const long rstep = (long)S * nw;
    long r = (long)s * nw + walker;
    for (; r + rstep < M; r += 2 * rstep) {
        long r2 = r + rstep;
        s16x8 vx0 = reinterpret_cast<const s16x8*>(x + r * C)[c8];
        s16x8 vg0 = reinterpret_cast<const s16x8*>(dy + r * C)[c8];
        s16x8 vx1 = reinterpret_cast<const s16x8*>(x + r2 * C)[c8];
        s16x8 vg1 = reinterpret_cast<const s16x8*>(dy + r2 * C)[c8];
        unsigned m0 = relu ? mask[r * cpgv + c8] : 0xffu;
        unsigned m1 = relu ? mask[r2 * cpgv + c8] : 0xffu;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float g = (m0 >> j) & 1u ? us2f((unsigned short)vg0[j]) : 0.f;
            float xh = (us2f((unsigned short)vx0[j]) - mean[j]) * invstd[j];
            sum_dy[j] += g;
            sum_dyx[j] = fmaf(g, xh, sum_dyx[j]);
            float g1 = (m1 >> j) & 1u ? us2f((unsigned short)vg1[j]) : 0.f;
            float xh1 = (us2f((unsigned short)vx1[j]) - mean[j]) * invstd[j];
            sum_dy[j] += g1;
            sum_dyx[j] = fmaf(g1, xh1, sum_dyx[j]);
        }
    }
    for (; r < M; r += rstep) {
        s16x8 vx = reinterpret_cast<const s16x8*>(x + r * C)[c8];
        s16x8 vg = reinterpret_cast<const s16x8*>(dy + r * C)[c8];
        unsigned m = relu ? mask[r * cpgv + c8] : 0xffu;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float g = (m >> j) & 1u ? us2f((unsigned short)vg[j]) : 0.f;
            float xh = (us2f((unsigned short)vx[j]) - mean[j]) * invstd[j];
            sum_dy[j] += g;
            sum_dyx[j] = fmaf(g, xh, sum_dyx[j]);
        }
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        red[walker * C + c8 * 8 + j] = sum_dy[j];
        red[nw * C + walker * C + c8 * 8 + j] = sum_dyx[j];
    }
    __syncthreads();
    for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float a = 0.f, b = 0.f;
        for (int w = 0; w < nw; ++w) {
            a += red[w * C + c];
            b += red[nw * C + w * C + c];
        }
        slab[(long)c * S + s] = a;
        slab[(long)C * S + (long)c * S + s] = b;
    }
}

__global__ void k_bn_bwd_combine(const float* __restrict__ slab, int S, int C,
                                 long M, const float* __restrict__ gamma,
                                 const float* __restrict__ save_invstd,
                                 float* __restrict__ dgamma,
                                 float* __restrict__ dbeta,
                                 float* __restrict__ coef_a,
                                 float* __restrict__ coef_b,
                                 float* __restrict__ coef_c) {
    int c = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    if (c >= C) return;
    float sum_dy = 0.f, sum_dyx = 0.f;
    for (int s = lane; s < S; s += 64) {
        sum_dy += slab[(long)c * S + s];
        sum_dyx += slab[(long)C * S + (long)c * S + s];
    }
    sum_dy = wave_reduce_sum(sum_dy);
    sum_dyx = wave_reduce_sum(sum_dyx);
    if (lane != 0) return;
    dgamma[c] = sum_dyx;
    dbeta[c] = sum_dy;
    float g = gamma ? gamma[c] : 1.f;
    coef_a[c] = g * save_invstd[c];
    coef_b[c] = sum_dy / M;
    coef_c[c] = sum_dyx / M;
}

// dx = a[c] * (dy_eff - b[c] - xhat * c[c]) — bf16x8.  Threads own a fixed
// 8-channel group (like the partial pass) so the five per-channel
// coefficients live in registers, not per-element loads.
__global__ void k_bn_bwd_dx(const bf16* __restrict__ x,
                            const bf16* __restrict__ dy,
                            const unsigned char* __restrict__ mask,
                            const float* __restrict__ save_mean,
                            const float* __restrict__ save_invstd,
                            const float* __restrict__ coef_a,
                            const float* __restrict__ coef_b,
                            const float* __restrict__ coef_c,
                            bf16* __restrict__ dx,
                            long M, int C, bool relu,
                            bf16* __restrict__ dresid) {
    const int cpg = C >> 3;
    const int c8 = threadIdx.x % cpg;
    const int walker = threadIdx.x / cpg;
    const int nw = blockDim.x / cpg;
    float mean[8], invstd[8], ca[8], cb[8], cc[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        int c = c8 * 8 + j;
        mean[j] = save_mean[c];
        invstd[j] = save_invstd[c];
        ca[j] = coef_a[c];
        cb[j] = coef_b[c];
        cc[j] = coef_c[c];
    }
    long r0 = (long)blockIdx.x * nw + walker;
    long rstride = (long)gridDim.x * nw;
    long r = r0;
    for (; r + rstride < M; r += 2 * rstride) {
        long r2 = r + rstride;
        s16x8 vx0 = reinterpret_cast<const s16x8*>(x + r * C)[c8];
        s16x8 vg0 = reinterpret_cast<const s16x8*>(dy + r * C)[c8];
        s16x8 vx1 = reinterpret_cast<const s16x8*>(x + r2 * C)[c8];
        s16x8 vg1 = reinterpret_cast<const s16x8*>(dy + r2 * C)[c8];
        unsigned m0 = relu ? mask[r * cpg + c8] : 0xffu;
        unsigned m1 = relu ? mask[r2 * cpg + c8] : 0xffu;
        s16x8 o0, og0, o1, og1;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float g = (m0 >> j) & 1u ? us2f((unsigned short)vg0[j]) : 0.f;
            float xh = (us2f((unsigned short)vx0[j]) - mean[j]) * invstd[j];
            o0[j] = (short)f2us(ca[j] * (g - cb[j] - xh * cc[j]));
            og0[j] = (short)f2us(g);
            float g1 = (m1 >> j) & 1u ? us2f((unsigned short)vg1[j]) : 0.f;
            float xh1 = (us2f((unsigned short)vx1[j]) - mean[j]) * invstd[j];
            o1[j] = (short)f2us(ca[j] * (g1 - cb[j] - xh1 * cc[j]));
            og1[j] = (short)f2us(g1);
        }
        reinterpret_cast<s16x8*>(dx + r * C)[c8] = o0;
        reinterpret_cast<s16x8*>(dx + r2 * C)[c8] = o1;
        if (dresid) {
            reinterpret_cast<s16x8*>(dresid + r * C)[c8] = og0;
            reinterpret_cast<s16x8*>(dresid + r2 * C)[c8] = og1;
        }
    }
    for (; r < M; r += rstride) {
        s16x8 vx = reinterpret_cast<const s16x8*>(x + r * C)[c8];
        s16x8 vg = reinterpret_cast<const s16x8*>(dy + r * C)[c8];
        unsigned m = relu ? mask[r * cpg + c8] : 0xffu;
        s16x8 o, og;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float g = (m >> j) & 1u ? us2f((unsigned short)vg[j]) : 0.f;
            float xh = (us2f((unsigned short)vx[j]) - mean[j]) * invstd[j];
            o[j] = (short)f2us(ca[j] * (g - cb[j] - xh * cc[j]));
            og[j] = (short)f2us(g);
        }
        reinterpret_cast<s16x8*>(dx + r * C)[c8] = o;
        if (dresid)   // skip-connection grad = relu-masked dy
            reinterpret_cast<s16x8*>(dresid + r * C)[c8] = og;
    }
}

// ----------------------------------------------- single-pass backward ------
//
// DDPX_BN1PASS=1: one persistent-grid kernel
// replaces partial+combine+dx.  Each workgroup computes its split's
// partial sums (phase 1, identical row striding and arithmetic to
// k_bn_bwd_partial), takes a ticket; the LAST workgroup combines the slab
// exactly like k_bn_bwd_combine (same lane/stride order — bitwise-equal
// coefficients), publishes them with a release flag; everyone else spins
// (s_sleep + acquire load), then runs the dx sweep over the SAME rows it
// just read — x/dy/mask stay hot in cache, saving one full read of each.
// The grid is sized to guaranteed-resident blocks
// (hipOccupancyMaxActiveBlocksPerMultiprocessor × CUs) so the ticket
// rendezvous cannot deadlock; the spin also carries a bail-out bound so a
// protocol bug degrades to wrong numbers (caught by the numerics tests),
// never a hung device.

__global__ __launch_bounds__(256)
void k_bn_bwd_onepass(const bf16* __restrict__ x,
                      const bf16* __restrict__ dy,
                      const unsigned char* __restrict__ mask,
                      const float* __restrict__ save_mean,
                      const float* __restrict__ save_invstd,
                      const float* __restrict__ gamma,
                      long M, int C, int S, bool relu,
                      float* __restrict__ slab,
                      float* __restrict__ dgamma,
                      float* __restrict__ dbeta,
                      float* __restrict__ coef_a,
                      float* __restrict__ coef_b,
                      float* __restrict__ coef_c,
                      int* __restrict__ ticket,
                      int* __restrict__ flag,
                      bf16* __restrict__ dx,
                      bf16* __restrict__ dresid) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* red = reinterpret_cast<float*>(smem);
    const int cpg = C >> 3;
    const int c8 = threadIdx.x % cpg;
    const int walker = threadIdx.x / cpg;
    const int nw = blockDim.x / cpg;
    const int s = blockIdx.x;
    float mean[8], invstd[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        mean[j] = save_mean[c8 * 8 + j];
        invstd[j] = save_invstd[c8 * 8 + j];
    }
    // ---- phase 1: partial sums over this split's rows (== k_bn_bwd_partial)
    float sum_dy[8] = {}, sum_dyx[8] = {};
    const long rstep = (long)S * nw;
    const long r0 = (long)s * nw + walker;
    for (long r = r0; r < M; r += rstep) {
        s16x8 vx = reinterpret_cast<const s16x8*>(x + r * C)[c8];
        s16x8 vg = reinterpret_cast<const s16x8*>(dy + r * C)[c8];
        unsigned m = relu ? mask[r * cpg + c8] : 0xffu;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float g = (m >> j) & 1u ? us2f((unsigned short)vg[j]) : 0.f;
            float xh = (us2f((unsigned short)vx[j]) - mean[j]) * invstd[j];
            sum_dy[j] += g;
            sum_dyx[j] = fmaf(g, xh, sum_dyx[j]);
        }
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        red[walker * C + c8 * 8 + j] = sum_dy[j];
        red[nw * C + walker * C + c8 * 8 + j] = sum_dyx[j];
    }
    __syncthreads();
    for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float a = 0.f, b = 0.f;
        for (int w = 0; w < nw; ++w) {
            a += red[w * C + c];
            b += red[nw * C + w * C + c];
        }
        slab[(long)c * S + s] = a;
        slab[(long)C * S + (long)c * S + s] = b;
    }
    // ---- ticket rendezvous
    __shared__ int is_last;
    __threadfence();                       // slab visible device-wide
    if (threadIdx.x == 0)
        is_last = (atomicAdd(ticket, 1) == gridDim.x - 1);
    __syncthreads();
    if (is_last) {
        // combine like k_bn_bwd_combine (wave per channel, lanes stride
        // the splits).  The slab lines were written by blocks on OTHER
        // XCDs: read them with cache-bypassing (relaxed agent-scope
        // atomic) loads — an acquire FENCE here would invalidate this
        // XCD's whole L2 and destroy the phase-2 cache-reuse premise.
        const int lane = threadIdx.x & 63;
        for (int c = (int)(threadIdx.x >> 6); c < C; c += (int)(blockDim.x >> 6)) {
            float a = 0.f, b = 0.f;
            for (int sp = lane; sp < S; sp += 64) {
                a += __hip_atomic_load(&slab[(long)c * S + sp],
                                       __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
                b += __hip_atomic_load(&slab[(long)C * S + (long)c * S + sp],
                                       __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
            }
            a = wave_reduce_sum(a);
            b = wave_reduce_sum(b);
            if (lane == 0) {
                dgamma[c] = b;
                dbeta[c] = a;
                float g = gamma ? gamma[c] : 1.f;
                coef_a[c] = g * save_invstd[c];
                coef_b[c] = a / M;
                coef_c[c] = b / M;
            }
        }
        __syncthreads();
        __threadfence();
        if (threadIdx.x == 0)
            __hip_atomic_store(flag, 1, __ATOMIC_RELEASE,
                               __HIP_MEMORY_SCOPE_AGENT);
    }
    // ---- spin for the coefficients (bounded: bail-out, never a hang).
    // RELAXED polls: an acquire load per poll emits an L2 invalidate —
    // hundreds of blocks polling would storm every XCD's L2 (measured:
    // ~20 ms per call).  A relaxed agent-scope atomic load bypasses the
    // stale cache without invalidating anything.
    if (threadIdx.x == 0) {
        long spins = 0;
        while (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_AGENT) == 0) {
            if (spins < 4) __builtin_amdgcn_s_sleep(2);
            else __builtin_amdgcn_s_sleep(64);
            if (++spins > (1l << 26)) break;   // ~seconds; wrong > hung
        }
    }
    __syncthreads();
    // ---- phase 2: dx over the SAME rows (cache-hot x/dy/mask).  The
    // coefficients come from the last block (another XCD): bypassing
    // loads again, NOT a fence — x/dy/mask must stay in cache.
    float ca[8], cb[8], cc[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        int c = c8 * 8 + j;
        ca[j] = __hip_atomic_load(&coef_a[c], __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
        cb[j] = __hip_atomic_load(&coef_b[c], __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
        cc[j] = __hip_atomic_load(&coef_c[c], __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
    }
    for (long r = r0; r < M; r += rstep) {
        s16x8 vx = reinterpret_cast<const s16x8*>(x + r * C)[c8];
        s16x8 vg = reinterpret_cast<const s16x8*>(dy + r * C)[c8];
        unsigned m = relu ? mask[r * cpg + c8] : 0xffu;
        s16x8 o, og;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float g = (m >> j) & 1u ? us2f((unsigned short)vg[j]) : 0.f;
            float xh = (us2f((unsigned short)vx[j]) - mean[j]) * invstd[j];
            o[j] = (short)f2us(ca[j] * (g - cb[j] - xh * cc[j]));
            og[j] = (short)f2us(g);
        }
        reinterpret_cast<s16x8*>(dx + r * C)[c8] = o;
        if (dresid)
            reinterpret_cast<s16x8*>(dresid + r * C)[c8] = og;
    }
}

// ------------------------------------------------------------------ hosts ---

// Guaranteed-resident grid for the one-pass rendezvous: blocks that are
// not simultaneously resident would deadlock the ticket protocol, so the
// grid is CUs × occupancy-per-CU, never more.
static int resident_grid(const void* kfunc, int block, int lds) {
    static int numCU = 0;
    if (!numCU) {
        hipDeviceProp_t p;
        if (hipGetDeviceProperties(&p, 0) != hipSuccess) return 0;
        numCU = p.multiProcessorCount;
    }
    // cache per (block, lds): the occupancy query does a kernel lookup
    // every call and BN backward runs ~50x per ResNet-50 step
    static std::unordered_map<long long, int> cache;
    static std::mutex mu;
    long long key = ((long long)block << 32) | (unsigned)lds;
    {
        std::lock_guard<std::mutex> lk(mu);
        auto it = cache.find(key);
        if (it != cache.end()) return it->second;
    }
    int per = 0;
    int grid = 0;
    if (hipOccupancyMaxActiveBlocksPerMultiprocessor(
            &per, kfunc, block, lds) == hipSuccess && per >= 1)
        grid = numCU * per;
    std::lock_guard<std::mutex> lk(mu);
    cache[key] = grid;
    return grid;
}

static void shape_mc(const at::Tensor& x, long& M, int& C) {
    if (x.dim() == 4) {        // NCHW logical, channels_last physical
        M = (long)x.size(0) * x.size(2) * x.size(3);
        C = x.size(1);
    } else {
        TORCH_CHECK(x.dim() == 2);
        M = x.size(0);
        C = x.size(1);
    }
}

static int pick_block(int C) {
    // threads = largest multiple of (C/8) channel-groups <= 256
    int cpg = C / 8;
    return (256 / cpg) * cpg;
}

std::vector<at::Tensor> bn_fwd_train(at::Tensor x, at::Tensor gamma,
                                     at::Tensor beta, at::Tensor running_mean,
                                     at::Tensor running_var, double momentum,
                                     double eps, bool fuse_relu,
                                     c10::optional<at::Tensor> residual,
                                     c10::optional<at::Tensor> pre_stats) {
    long M; int C;
    shape_mc(x, M, C);
    TORCH_CHECK(C % 8 == 0 && C / 8 <= 256, "bn HIP path needs C%8==0");
    auto stream = at::hip::getCurrentHIPStream();
    auto fopts = gamma.options().dtype(at::kFloat);
    int block = pick_block(C);
    int nw = block / (C / 8);
    // pre_stats: [2][C][S] partial sums already produced by the conv
    // epilogue — the stats pass (a full re-read of x) is skipped.
    int S = pre_stats.has_value() ? (int)pre_stats->size(2)
                                  : bn_splits(M, nw);
    auto slab = pre_stats.has_value() ? *pre_stats
                                      : at::empty({2, C, S}, fopts);
    auto save_mean = at::empty({C}, fopts);
    auto save_invstd = at::empty({C}, fopts);
    auto scale = at::empty({C}, fopts);
    auto shift = at::empty({C}, fopts);
    auto y = x.dim() == 4
        ? at::empty_like(x, x.options().memory_format(at::MemoryFormat::ChannelsLast))
        : at::empty_like(x);
    const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());
    int lds = 2 * nw * C * 4;

    if (!pre_stats.has_value()) {
        hipLaunchKernelGGL(k_bn_partial, dim3(S), dim3(block), lds,
                           stream.stream(), xp, M, C, S,
                           slab.data_ptr<float>());
        HIP_CHECK_LAST();
    }
    hipLaunchKernelGGL(k_bn_combine, dim3(ceil_div_i(C, 4)), dim3(256), 0,
                       stream.stream(), slab.data_ptr<float>(), S, C, M,
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(),
                       (float)momentum, (float)eps,
                       save_mean.data_ptr<float>(),
                       save_invstd.data_ptr<float>(),
                       scale.data_ptr<float>(), shift.data_ptr<float>());
    HIP_CHECK_LAST();
    long tv = M * C / 8;
    bf16* yp = reinterpret_cast<bf16*>(y.data_ptr());
    int blocks = std::min<long>(4096, ceil_div_i(tv, 256));
    const bf16* rp = residual.has_value()
        ? reinterpret_cast<const bf16*>(residual->data_ptr()) : nullptr;
    at::Tensor mask;
    unsigned char* mp = nullptr;
    if (fuse_relu) {   // packed relu mask: byte per 8-channel group (bwd
        mask = at::empty({M, C / 8}, x.options().dtype(at::kByte));
        mp = mask.data_ptr<unsigned char>();   // reads 1/16 of y's bytes)
    } else {
        mask = at::empty({0}, x.options().dtype(at::kByte));
    }
    hipLaunchKernelGGL(k_bn_apply_v8, dim3(blocks), dim3(256), 0,
                       stream.stream(), xp, yp, scale.data_ptr<float>(),
                       shift.data_ptr<float>(), tv, C / 8, fuse_relu, rp, mp);
    HIP_CHECK_LAST();
    return {y, save_mean, save_invstd, mask};
}

// Eval forward that ALSO emits the packed relu mask (needed to backprop
// through a frozen/eval BN with fused ReLU — fine-tuning-style use;
// VERDICT r01 weak #8).
std::vector<at::Tensor> bn_fwd_eval_mask(at::Tensor x, at::Tensor gamma,
                                         at::Tensor beta,
                                         at::Tensor running_mean,
                                         at::Tensor running_var,
                                         double eps, bool fuse_relu,
                                         c10::optional<at::Tensor> residual) {
    long M; int C;
    shape_mc(x, M, C);
    TORCH_CHECK(C % 8 == 0);
    auto stream = at::hip::getCurrentHIPStream();
    auto fopts = gamma.options().dtype(at::kFloat);
    auto scale = at::empty({C}, fopts);
    auto shift = at::empty({C}, fopts);
    auto y = x.dim() == 4
        ? at::empty_like(x, x.options().memory_format(at::MemoryFormat::ChannelsLast))
        : at::empty_like(x);
    hipLaunchKernelGGL(k_bn_eval_coeffs, dim3(ceil_div_i(C, 256)), dim3(256), 0,
                       stream.stream(), gamma.data_ptr<float>(),
                       beta.data_ptr<float>(), running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(), (float)eps, C,
                       scale.data_ptr<float>(), shift.data_ptr<float>());
    HIP_CHECK_LAST();
    long tv = M * C / 8;
    int blocks = std::min<long>(4096, ceil_div_i(tv, 256));
    const bf16* rp = residual.has_value()
        ? reinterpret_cast<const bf16*>(residual->data_ptr()) : nullptr;
    at::Tensor mask;
    unsigned char* mp = nullptr;
    if (fuse_relu) {
        mask = at::empty({M, C / 8}, x.options().dtype(at::kByte));
        mp = mask.data_ptr<unsigned char>();
    } else {
        mask = at::empty({0}, x.options().dtype(at::kByte));
    }
    hipLaunchKernelGGL(k_bn_apply_v8, dim3(blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()),
                       scale.data_ptr<float>(), shift.data_ptr<float>(), tv,
                       C / 8, fuse_relu, rp, mp);
    HIP_CHECK_LAST();
    return {y, mask};
}

at::Tensor bn_fwd_eval(at::Tensor x, at::Tensor gamma, at::Tensor beta,
                       at::Tensor running_mean, at::Tensor running_var,
                       double eps, bool fuse_relu,
                       c10::optional<at::Tensor> residual) {
    long M; int C;
    shape_mc(x, M, C);
    TORCH_CHECK(C % 8 == 0);
    auto stream = at::hip::getCurrentHIPStream();
    auto fopts = gamma.options().dtype(at::kFloat);
    auto scale = at::empty({C}, fopts);
    auto shift = at::empty({C}, fopts);
    auto y = x.dim() == 4
        ? at::empty_like(x, x.options().memory_format(at::MemoryFormat::ChannelsLast))
        : at::empty_like(x);
    hipLaunchKernelGGL(k_bn_eval_coeffs, dim3(ceil_div_i(C, 256)), dim3(256), 0,
                       stream.stream(), gamma.data_ptr<float>(),
                       beta.data_ptr<float>(), running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(), (float)eps, C,
                       scale.data_ptr<float>(), shift.data_ptr<float>());
    HIP_CHECK_LAST();
    long tv = M * C / 8;
    int blocks = std::min<long>(4096, ceil_div_i(tv, 256));
    const bf16* rp = residual.has_value()
        ? reinterpret_cast<const bf16*>(residual->data_ptr()) : nullptr;
    hipLaunchKernelGGL(k_bn_apply_v8, dim3(blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()),
                       scale.data_ptr<float>(), shift.data_ptr<float>(), tv,
                       C / 8, fuse_relu, rp, nullptr);
    HIP_CHECK_LAST();
    return y;
}

std::vector<at::Tensor> bn_bwd(at::Tensor x, at::Tensor dy, at::Tensor gamma,
                               at::Tensor save_mean, at::Tensor save_invstd,
                               at::Tensor mask, bool fuse_relu,
                               bool want_dresid,
                               c10::optional<at::Tensor> pre_slab,
                               bool eval_stats) {
    long M; int C;
    shape_mc(x, M, C);
    TORCH_CHECK(C % 8 == 0 && C / 8 <= 256);
    auto stream = at::hip::getCurrentHIPStream();
    auto fopts = gamma.options().dtype(at::kFloat);
    int block = pick_block(C);
    int nw = block / (C / 8);
    // pre_slab: backward partials emitted by the downstream conv's dgrad
    // epilogue (producer-side fusion) — the partial pass is skipped.
    int S = pre_slab.has_value() ? (int)pre_slab->size(2)
                                 : bn_splits(M, nw);
    auto slab = pre_slab.has_value() ? *pre_slab
                                     : at::empty({2, C, S}, fopts);
    auto dgamma = at::empty({C}, fopts);
    auto dbeta = at::empty({C}, fopts);
    auto ca = at::empty({C}, fopts);
    auto cb = at::empty({C}, fopts);
    auto cc = at::empty({C}, fopts);
    auto dx = x.dim() == 4
        ? at::empty_like(x, x.options().memory_format(at::MemoryFormat::ChannelsLast))
        : at::empty_like(x);
    const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());
    const bf16* dyp = reinterpret_cast<const bf16*>(dy.data_ptr());
    TORCH_CHECK(!fuse_relu || mask.numel() == M * (C / 8),
                "bn_bwd needs the packed relu mask from bn_fwd_train");
    const unsigned char* yp = fuse_relu
        ? mask.data_ptr<unsigned char>() : nullptr;
    int lds = 2 * nw * C * 4;

    const char* e1p = getenv("DDPX_BN1PASS");
    if (e1p && e1p[0] == '1' && !pre_slab.has_value() && !eval_stats) {
        int grid = resident_grid((const void*)k_bn_bwd_onepass, block, lds);
        if (grid > 0) {
            // no empty splits: blocks with no rows still pay the full
            // rendezvous+spin, so cap the grid at the rows available
            int S1 = (int)std::min<long>(grid, ceil_div_i(M, nw));
            auto slab1 = at::empty({2, C, S1}, fopts);
            auto work = at::zeros({2}, gamma.options().dtype(at::kInt));
            at::Tensor dresid1;
            bf16* drp1 = nullptr;
            if (want_dresid) {
                dresid1 = x.dim() == 4
                    ? at::empty_like(x, x.options().memory_format(
                          at::MemoryFormat::ChannelsLast))
                    : at::empty_like(x);
                drp1 = reinterpret_cast<bf16*>(dresid1.data_ptr());
            }
            hipLaunchKernelGGL(k_bn_bwd_onepass, dim3(S1), dim3(block), lds,
                               stream.stream(), xp, dyp, yp,
                               save_mean.data_ptr<float>(),
                               save_invstd.data_ptr<float>(),
                               gamma.data_ptr<float>(), M, C, S1, fuse_relu,
                               slab1.data_ptr<float>(),
                               dgamma.data_ptr<float>(),
                               dbeta.data_ptr<float>(),
                               ca.data_ptr<float>(), cb.data_ptr<float>(),
                               cc.data_ptr<float>(),
                               work.data_ptr<int>(), work.data_ptr<int>() + 1,
                               reinterpret_cast<bf16*>(dx.data_ptr()), drp1);
            HIP_CHECK_LAST();
            if (want_dresid) return {dx, dgamma, dbeta, dresid1};
            return {dx, dgamma, dbeta};
        }
    }

    if (!pre_slab.has_value()) {
        hipLaunchKernelGGL(k_bn_bwd_partial, dim3(S), dim3(block), lds,
                           stream.stream(), xp, dyp, yp,
                           save_mean.data_ptr<float>(),
                           save_invstd.data_ptr<float>(), M, C, S, fuse_relu,
                           slab.data_ptr<float>());
        HIP_CHECK_LAST();
    }
    hipLaunchKernelGGL(k_bn_bwd_combine, dim3(ceil_div_i(C, 4)), dim3(256), 0,
                       stream.stream(), slab.data_ptr<float>(), S, C, M,
                       gamma.data_ptr<float>(), save_invstd.data_ptr<float>(),
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                       ca.data_ptr<float>(), cb.data_ptr<float>(),
                       cc.data_ptr<float>());
    HIP_CHECK_LAST();
    if (eval_stats) {
        // frozen statistics: mean/var are constants, so the batch
        // mean-correction terms vanish — dx = gamma*invstd * dy_eff
        cb.zero_();
        cc.zero_();
    }
    at::Tensor dresid;
    bf16* drp = nullptr;
    if (want_dresid) {
        dresid = x.dim() == 4
            ? at::empty_like(x, x.options().memory_format(at::MemoryFormat::ChannelsLast))
            : at::empty_like(x);
        drp = reinterpret_cast<bf16*>(dresid.data_ptr());
    }
    int dxblocks = (int)std::max<long>(1, std::min<long>(2048, M / (nw * 2)));
    hipLaunchKernelGGL(k_bn_bwd_dx, dim3(dxblocks), dim3(block), 0,
                       stream.stream(),
                       xp, dyp, yp, save_mean.data_ptr<float>(),
                       save_invstd.data_ptr<float>(), ca.data_ptr<float>(),
                       cb.data_ptr<float>(), cc.data_ptr<float>(),
                       reinterpret_cast<bf16*>(dx.data_ptr()), M, C,
                       fuse_relu, drp);
    HIP_CHECK_LAST();
    if (want_dresid) return {dx, dgamma, dbeta, dresid};
    return {dx, dgamma, dbeta};
}
