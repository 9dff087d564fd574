#include "hip/hip_runtime.h"
// bn.hip — BatchNorm (2d NHWC and 1d NC) forward/backward with fused ReLU
// epilogue, fp32 statistics over bf16 activations.
//
// Both cases reduce over M rows × C channels where M = N*H*W (2d,
// channels_last) or N (1d): lanes run along C → fully coalesced.
// Reductions are deterministic: S fixed split partials into a slab,
// fixed-order combine (no fp atomics) — bit-reproducible under
// same_seeds like the whole gradient path.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

constexpr int BN_CB = 64;    // channels per block
constexpr int BN_RW = 4;     // row-walkers per channel (block = 256 threads)

static inline int bn_splits(long M, int C) {
    // target >=1024 blocks for the partial pass, capped by row count
    int cb = ceil_div_i(C, BN_CB);
    int s = std::max(1, 1024 / std::max(cb, 1));
    s = std::min<long>(s, (M + BN_RW - 1) / BN_RW);
    return std::max(1, s);
}

// partial sums: slab[s][c] = {sum, sumsq} over rows s::S
__global__ void k_bn_partial(const bf16* __restrict__ x, long M, int C,
                             int S, float* __restrict__ slab) {
    __shared__ float red[2][BN_RW][BN_CB];
    int cb = blockIdx.x;           // channel block
    int s = blockIdx.y;            // split
    int c = cb * BN_CB + (threadIdx.x % BN_CB);
    int walker = threadIdx.x / BN_CB;
    float sum = 0.f, sq = 0.f;
    if (c < C) {
        for (long r = s * BN_RW + walker; r < M; r += (long)S * BN_RW) {
            float v = bf2f(x[r * C + c]);
            sum += v;
            sq = fmaf(v, v, sq);
        }
    }
    red[0][walker][threadIdx.x % BN_CB] = sum;
    red[1][walker][threadIdx.x % BN_CB] = sq;
    __syncthreads();
    if (walker == 0 && c < C) {
        float ts = 0.f, tq = 0.f;
        #pragma unroll
        for (int w = 0; w < BN_RW; ++w) {
            ts += red[0][w][threadIdx.x % BN_CB];
            tq += red[1][w][threadIdx.x % BN_CB];
        }
        slab[((long)s * C + c) * 2 + 0] = ts;
        slab[((long)s * C + c) * 2 + 1] = tq;
    }
}

// combine: per-channel stats, scale/shift, running-stat update
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
    int c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float sum = 0.f, sq = 0.f;
    for (int s = 0; s < S; ++s) {          // fixed order: deterministic
        sum += slab[((long)s * C + c) * 2 + 0];
        sq += slab[((long)s * C + c) * 2 + 1];
    }
    float mean = sum / M;
    float var = fmaxf(sq / M - mean * mean, 0.f);   // biased (normalization)
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

// eval-mode scale/shift from running stats
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

// apply: y = relu?(x*scale[c] + shift[c]) — vectorized bf16x8, C % 8 == 0 path
__global__ void k_bn_apply_v8(const bf16* __restrict__ x, bf16* __restrict__ y,
                              const float* __restrict__ scale,
                              const float* __restrict__ shift,
                              long total_v, int Cv, bool relu) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total_v; i += stride) {
        int cv = (i % Cv) * 8;
        s16x8 v = reinterpret_cast<const s16x8*>(x)[i];
        s16x8 o;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = fmaf(us2f((unsigned short)v[j]), scale[cv + j], shift[cv + j]);
            if (relu) f = fmaxf(f, 0.f);
            o[j] = (short)f2us(f);
        }
        reinterpret_cast<s16x8*>(y)[i] = o;
    }
}

__global__ void k_bn_apply_scalar(const bf16* __restrict__ x, bf16* __restrict__ y,
                                  const float* __restrict__ scale,
                                  const float* __restrict__ shift,
                                  long total, int C, bool relu) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int c = i % C;
        float f = fmaf(bf2f(x[i]), scale[c], shift[c]);
        if (relu) f = fmaxf(f, 0.f);
        y[i] = f2bf(f);
    }
}

// backward partials: per-channel {sum(dy_eff), sum(dy_eff * xhat)}
__global__ void k_bn_bwd_partial(const bf16* __restrict__ x,
                                 const bf16* __restrict__ dy,
                                 const bf16* __restrict__ y,  // post-relu (mask)
                                 const float* __restrict__ save_mean,
                                 const float* __restrict__ save_invstd,
                                 long M, int C, int S, bool relu,
                                 float* __restrict__ slab) {
    __shared__ float red[2][BN_RW][BN_CB];
    int cb = blockIdx.x;
    int s = blockIdx.y;
    int c = cb * BN_CB + (threadIdx.x % BN_CB);
    int walker = threadIdx.x / BN_CB;
    float sum_dy = 0.f, sum_dyx = 0.f;
    if (c < C) {
        float mean = save_mean[c], invstd = save_invstd[c];
        for (long r = s * BN_RW + walker; r < M; r += (long)S * BN_RW) {
            float g = bf2f(dy[r * C + c]);
            if (relu && bf2f(y[r * C + c]) <= 0.f) g = 0.f;
            float xh = (bf2f(x[r * C + c]) - mean) * invstd;
            sum_dy += g;
            sum_dyx = fmaf(g, xh, sum_dyx);
        }
    }
    red[0][walker][threadIdx.x % BN_CB] = sum_dy;
    red[1][walker][threadIdx.x % BN_CB] = sum_dyx;
    __syncthreads();
    if (walker == 0 && c < C) {
        float a = 0.f, b = 0.f;
        #pragma unroll
        for (int w = 0; w < BN_RW; ++w) {
            a += red[0][w][threadIdx.x % BN_CB];
            b += red[1][w][threadIdx.x % BN_CB];
        }
        slab[((long)s * C + c) * 2 + 0] = a;
        slab[((long)s * C + c) * 2 + 1] = b;
    }
}

// combine backward: dgamma, dbeta + dx coefficients
__global__ void k_bn_bwd_combine(const float* __restrict__ slab, int S, int C,
                                 long M, const float* __restrict__ gamma,
                                 const float* __restrict__ save_invstd,
                                 float* __restrict__ dgamma,
                                 float* __restrict__ dbeta,
                                 float* __restrict__ coef_a,  // gamma*invstd
                                 float* __restrict__ coef_b,  // dbeta/M
                                 float* __restrict__ coef_c)  // sum_dyx/M
{
    int c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float sum_dy = 0.f, sum_dyx = 0.f;
    for (int s = 0; s < S; ++s) {
        sum_dy += slab[((long)s * C + c) * 2 + 0];
        sum_dyx += slab[((long)s * C + c) * 2 + 1];
    }
    dgamma[c] = sum_dyx;
    dbeta[c] = sum_dy;
    float g = gamma ? gamma[c] : 1.f;
    coef_a[c] = g * save_invstd[c];
    coef_b[c] = sum_dy / M;
    coef_c[c] = sum_dyx / M;
}

// dx = a[c] * (dy_eff - b[c] - xhat * c[c])
__global__ void k_bn_bwd_dx(const bf16* __restrict__ x,
                            const bf16* __restrict__ dy,
                            const bf16* __restrict__ y,
                            const float* __restrict__ save_mean,
                            const float* __restrict__ save_invstd,
                            const float* __restrict__ coef_a,
                            const float* __restrict__ coef_b,
                            const float* __restrict__ coef_c,
                            bf16* __restrict__ dx,
                            long total, int C, bool relu) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int c = i % C;
        float g = bf2f(dy[i]);
        if (relu && bf2f(y[i]) <= 0.f) g = 0.f;
        float xh = (bf2f(x[i]) - save_mean[c]) * save_invstd[c];
        dx[i] = f2bf(coef_a[c] * (g - coef_b[c] - xh * coef_c[c]));
    }
}

// ------------------------------------------------------------------ hosts ---

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

std::vector<at::Tensor> bn_fwd_train(at::Tensor x, at::Tensor gamma,
                                     at::Tensor beta, at::Tensor running_mean,
                                     at::Tensor running_var, double momentum,
                                     double eps, bool fuse_relu) {
    long M; int C;
    shape_mc(x, M, C);
    auto stream = at::hip::getCurrentHIPStream();
    auto fopts = gamma.options().dtype(at::kFloat);
    int S = bn_splits(M, C);
    auto slab = at::empty({S, C, 2}, fopts);
    auto save_mean = at::empty({C}, fopts);
    auto save_invstd = at::empty({C}, fopts);
    auto scale = at::empty({C}, fopts);
    auto shift = at::empty({C}, fopts);
    auto y = x.dim() == 4
        ? at::empty_like(x, x.options().memory_format(at::MemoryFormat::ChannelsLast))
        : at::empty_like(x);
    const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());

    dim3 pgrid(ceil_div_i(C, BN_CB), S);
    hipLaunchKernelGGL(k_bn_partial, pgrid, dim3(BN_CB * BN_RW), 0,
                       stream.stream(), xp, M, C, S, slab.data_ptr<float>());
    HIP_CHECK_LAST();
    hipLaunchKernelGGL(k_bn_combine, dim3(ceil_div_i(C, 256)), dim3(256), 0,
                       stream.stream(), slab.data_ptr<float>(), S, C, M,
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(),
                       (float)momentum, (float)eps,
                       save_mean.data_ptr<float>(),
                       save_invstd.data_ptr<float>(),
                       scale.data_ptr<float>(), shift.data_ptr<float>());
    HIP_CHECK_LAST();
    long total = M * C;
    bf16* yp = reinterpret_cast<bf16*>(y.data_ptr());
    if (C % 8 == 0) {
        long tv = total / 8;
        int blocks = std::min<long>(4096, ceil_div_i(tv, 256));
        hipLaunchKernelGGL(k_bn_apply_v8, dim3(blocks), dim3(256), 0,
                           stream.stream(), xp, yp, scale.data_ptr<float>(),
                           shift.data_ptr<float>(), tv, C / 8, fuse_relu);
    } else {
        int blocks = std::min<long>(4096, ceil_div_i(total, 256));
        hipLaunchKernelGGL(k_bn_apply_scalar, dim3(blocks), dim3(256), 0,
                           stream.stream(), xp, yp, scale.data_ptr<float>(),
                           shift.data_ptr<float>(), total, C, fuse_relu);
    }
    HIP_CHECK_LAST();
    return {y, save_mean, save_invstd};
}

at::Tensor bn_fwd_eval(at::Tensor x, at::Tensor gamma, at::Tensor beta,
                       at::Tensor running_mean, at::Tensor running_var,
                       double eps, bool fuse_relu) {
    long M; int C;
    shape_mc(x, M, C);
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
    const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());
    bf16* yp = reinterpret_cast<bf16*>(y.data_ptr());
    long total = M * C;
    if (C % 8 == 0) {
        long tv = total / 8;
        int blocks = std::min<long>(4096, ceil_div_i(tv, 256));
        hipLaunchKernelGGL(k_bn_apply_v8, dim3(blocks), dim3(256), 0,
                           stream.stream(), xp, yp, scale.data_ptr<float>(),
                           shift.data_ptr<float>(), tv, C / 8, fuse_relu);
    } else {
        int blocks = std::min<long>(4096, ceil_div_i(total, 256));
        hipLaunchKernelGGL(k_bn_apply_scalar, dim3(blocks), dim3(256), 0,
                           stream.stream(), xp, yp, scale.data_ptr<float>(),
                           shift.data_ptr<float>(), total, C, fuse_relu);
    }
    HIP_CHECK_LAST();
    return y;
}

std::vector<at::Tensor> bn_bwd(at::Tensor x, at::Tensor dy, at::Tensor gamma,
                               at::Tensor save_mean, at::Tensor save_invstd,
                               at::Tensor y, bool fuse_relu) {
    long M; int C;
    shape_mc(x, M, C);
    auto stream = at::hip::getCurrentHIPStream();
    auto fopts = gamma.options().dtype(at::kFloat);
    int S = bn_splits(M, C);
    auto slab = at::empty({S, C, 2}, fopts);
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
    const bf16* yp = reinterpret_cast<const bf16*>(y.data_ptr());

    dim3 pgrid(ceil_div_i(C, BN_CB), S);
    hipLaunchKernelGGL(k_bn_bwd_partial, pgrid, dim3(BN_CB * BN_RW), 0,
                       stream.stream(), xp, dyp, yp,
                       save_mean.data_ptr<float>(),
                       save_invstd.data_ptr<float>(), M, C, S, fuse_relu,
                       slab.data_ptr<float>());
    HIP_CHECK_LAST();
    hipLaunchKernelGGL(k_bn_bwd_combine, dim3(ceil_div_i(C, 256)), dim3(256), 0,
                       stream.stream(), slab.data_ptr<float>(), S, C, M,
                       gamma.data_ptr<float>(), save_invstd.data_ptr<float>(),
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                       ca.data_ptr<float>(), cb.data_ptr<float>(),
                       cc.data_ptr<float>());
    HIP_CHECK_LAST();
    long total = M * C;
    int blocks = std::min<long>(4096, ceil_div_i(total, 256));
    hipLaunchKernelGGL(k_bn_bwd_dx, dim3(blocks), dim3(256), 0, stream.stream(),
                       xp, dyp, yp, save_mean.data_ptr<float>(),
                       save_invstd.data_ptr<float>(), ca.data_ptr<float>(),
                       cb.data_ptr<float>(), cc.data_ptr<float>(),
                       reinterpret_cast<bf16*>(dx.data_ptr()), total, C,
                       fuse_relu);
    HIP_CHECK_LAST();
    return {dx, dgamma, dbeta};
}
