// common.h — shared helpers for the gfx950 (CDNA4) kernels.
//
// Conventions:
//  * wave size 64 (CDNA), block sizes multiples of 64;
//  * bf16 activations/weights, fp32 accumulation and statistics;
//  * all reductions on the gradient path are fixed-order (slab partials +
//    deterministic combine) — no fp atomics — so training is bit-reproducible
//    (replaces the reference's cudnn.deterministic contract);
//  * grids sized ≫256 workgroups where the problem allows (8 XCDs × 32 CUs).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEV_INLINE __device__ __forceinline__

typedef __hip_bfloat16 bf16;

// vector types for wide loads/stores
typedef short  s16x2 __attribute__((ext_vector_type(2)));
typedef short  s16x4 __attribute__((ext_vector_type(4)));
typedef short  s16x8 __attribute__((ext_vector_type(8)));
typedef float  f32x2 __attribute__((ext_vector_type(2)));
typedef float  f32x4 __attribute__((ext_vector_type(4)));
typedef float  f32x16 __attribute__((ext_vector_type(16)));

constexpr int WAVE = 64;

DEV_INLINE float bf2f(bf16 v) { return __bfloat162float(v); }
DEV_INLINE bf16 f2bf(float v) { return __float2bfloat16(v); }

// bf16 stored as ushort bit pattern helpers (for vectorized paths)
DEV_INLINE float us2f(unsigned short u) {
    union { unsigned int i; float f; } c;
    c.i = ((unsigned int)u) << 16;
    return c.f;
}
DEV_INLINE unsigned short f2us(float f) {
    union { unsigned int i; float f; } c;
    c.f = f;
    unsigned int i = c.i;
    // round-to-nearest-even like __float2bfloat16
    unsigned int lsb = (i >> 16) & 1u;
    i += 0x7fffu + lsb;
    return (unsigned short)(i >> 16);
}

// wave-wide reductions (64 lanes)
DEV_INLINE float wave_reduce_sum(float v) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, 64);
    return v;  // valid in lane 0
}
DEV_INLINE float wave_reduce_max(float v) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_down(v, off, 64));
    return v;
}

DEV_INLINE long long lds_block_reduce_ll(long long v, long long* scratch) {
    // block-level sum of per-wave partials (blockDim.x threads, <=16 waves)
    int lane = threadIdx.x & 63;
    int wid = threadIdx.x >> 6;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, 64);
    if (lane == 0) scratch[wid] = v;
    __syncthreads();
    int nw = (blockDim.x + 63) >> 6;
    long long out = 0;
    if (threadIdx.x == 0) {
        for (int i = 0; i < nw; ++i) out += scratch[i];
    }
    return out;  // valid in thread 0
}

// Fast unsigned division by a runtime constant (libdivide-style
// multiply-shift, set up once on the host).  The conv gathers decode
// (m -> n,p,q) and (k -> r,s,c) per staged element; s_div emulation costs
// tens of VALU cycles each, v_mul_hi_u32 + shift costs ~4.
// Powers of two (incl. 1) use a plain shift (mul == 0 flags it); otherwise
// the Granlund–Montgomery round-up form: shift = floor(log2 d),
// mul = floor(2^(32+shift)/d) + 1, q = (n*mul) >> (32+shift) — exact for
// n < 2^31 (all index spaces here are far below that).  d is wave-uniform
// so the mul==0 branch is a scalar branch.
struct FastDiv {
    unsigned mul;
    unsigned shift;
    unsigned d;
    __host__ void init(unsigned div) {
        d = div;
        if ((div & (div - 1)) == 0) {   // power of two (incl. 1)
            mul = 0;
            shift = 0;
            while ((1u << shift) < div) ++shift;
            return;
        }
        shift = 0;
        while ((2u << shift) < div) ++shift;   // floor(log2(div))
        mul = (unsigned)(((1ull << (32 + shift)) / div) + 1);
    }
};
DEV_INLINE unsigned fd_div(unsigned n, FastDiv f) {
    if (f.mul == 0) return n >> f.shift;
    return (unsigned)(((unsigned long long)n * f.mul) >> 32) >> f.shift;
}
DEV_INLINE unsigned fd_mod(unsigned n, FastDiv f, unsigned q) {
    return n - q * f.d;
}

#define HIP_CHECK_LAST()                                                     \
    do {                                                                     \
        hipError_t e = hipGetLastError();                                    \
        if (e != hipSuccess) {                                               \
            TORCH_CHECK(false, "HIP kernel launch failed: ",                 \
                        hipGetErrorString(e));                               \
        }                                                                    \
    } while (0)

static inline int ceil_div_i(long long a, long long b) {
    return (int)((a + b - 1) / b);
}
