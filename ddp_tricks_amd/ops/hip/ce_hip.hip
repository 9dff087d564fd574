#include "hip/hip_runtime.h"
// ce.hip — fused cross-entropy (log-softmax + NLL) fwd/bwd and the
// rowwise-argmax accuracy reduction.
//
// Row length is small (10 classes for the flagship config) so each thread
// owns one row; the loss mean over the batch is a deterministic two-stage
// reduction (per-block partials in fixed order, single-block combine) —
// no fp atomics.  Backward reads the upstream gradient from device memory
// (0-dim tensor) so no host sync is needed.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

constexpr int CE_BLOCK = 256;

// per-row: lse_i = log(sum exp(x - max)) + max ; rowloss_i = lse_i - x[t_i]
template <typename T>
__global__ void k_ce_rows(const T* __restrict__ logits,
                          const long* __restrict__ target,
                          float* __restrict__ rowloss,
                          float* __restrict__ lse_out,
                          int B, int C) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= B) return;
    const T* row = logits + (long)i * C;
    float mx = -INFINITY;
    for (int c = 0; c < C; ++c) mx = fmaxf(mx, (float)row[c]);
    float s = 0.f;
    for (int c = 0; c < C; ++c) s += expf((float)row[c] - mx);
    float lse = logf(s) + mx;
    lse_out[i] = lse;
    rowloss[i] = lse - (float)row[target[i]];
}

// deterministic mean: single block, fixed-order tree over per-thread partials
__global__ void k_reduce_mean(const float* __restrict__ rowloss, int B,
                              float* __restrict__ out) {
    __shared__ float partial[CE_BLOCK];
    float acc = 0.f;
    for (int i = threadIdx.x; i < B; i += CE_BLOCK) acc += rowloss[i];
    partial[threadIdx.x] = acc;
    __syncthreads();
    for (int off = CE_BLOCK / 2; off > 0; off >>= 1) {
        if (threadIdx.x < off) partial[threadIdx.x] += partial[threadIdx.x + off];
        __syncthreads();
    }
    if (threadIdx.x == 0) out[0] = partial[0] / B;
}

std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor target) {
    TORCH_CHECK(logits.dim() == 2 && logits.is_cuda() && logits.is_contiguous());
    TORCH_CHECK(target.scalar_type() == at::kLong);
    int B = logits.size(0), C = logits.size(1);
    auto opts = logits.options().dtype(at::kFloat);
    auto rowloss = at::empty({B}, opts);
    auto lse = at::empty({B}, opts);
    auto loss = at::empty({}, opts);
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = ceil_div_i(B, CE_BLOCK);
    if (logits.scalar_type() == at::kBFloat16) {
        hipLaunchKernelGGL(k_ce_rows<bf16>, dim3(blocks), dim3(CE_BLOCK), 0,
                           stream.stream(),
                           reinterpret_cast<const bf16*>(logits.data_ptr()),
                           target.data_ptr<long>(), rowloss.data_ptr<float>(),
                           lse.data_ptr<float>(), B, C);
    } else {
        TORCH_CHECK(logits.scalar_type() == at::kFloat);
        hipLaunchKernelGGL(k_ce_rows<float>, dim3(blocks), dim3(CE_BLOCK), 0,
                           stream.stream(), logits.data_ptr<float>(),
                           target.data_ptr<long>(), rowloss.data_ptr<float>(),
                           lse.data_ptr<float>(), B, C);
    }
    HIP_CHECK_LAST();
    hipLaunchKernelGGL(k_reduce_mean, dim3(1), dim3(CE_BLOCK), 0,
                       stream.stream(), rowloss.data_ptr<float>(),
                       B, loss.data_ptr<float>());
    HIP_CHECK_LAST();
    return {loss, lse};
}

// dlogits[i,c] = (softmax(i,c) - [c==t_i]) * (*dloss) / B
template <typename T>
__global__ void k_ce_bwd(const T* __restrict__ logits,
                         const long* __restrict__ target,
                         const float* __restrict__ lse,
                         const float* __restrict__ dloss,
                         T* __restrict__ dlogits, int B, int C) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= B) return;
    float g = dloss[0] / B;
    const T* row = logits + (long)i * C;
    T* drow = dlogits + (long)i * C;
    float l = lse[i];
    long t = target[i];
    for (int c = 0; c < C; ++c) {
        float p = expf((float)row[c] - l);
        drow[c] = (T)((p - (c == t ? 1.f : 0.f)) * g);
    }
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor target, at::Tensor lse,
                  at::Tensor dloss) {
    int B = logits.size(0), C = logits.size(1);
    auto dlogits = at::empty_like(logits);
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = ceil_div_i(B, CE_BLOCK);
    auto dlossf = dloss.scalar_type() == at::kFloat ? dloss : dloss.to(at::kFloat);
    if (logits.scalar_type() == at::kBFloat16) {
        hipLaunchKernelGGL(k_ce_bwd<bf16>, dim3(blocks), dim3(CE_BLOCK), 0,
                           stream.stream(),
                           reinterpret_cast<const bf16*>(logits.data_ptr()),
                           target.data_ptr<long>(), lse.data_ptr<float>(),
                           dlossf.data_ptr<float>(),
                           reinterpret_cast<bf16*>(dlogits.data_ptr()), B, C);
    } else {
        hipLaunchKernelGGL(k_ce_bwd<float>, dim3(blocks), dim3(CE_BLOCK), 0,
                           stream.stream(), logits.data_ptr<float>(),
                           target.data_ptr<long>(), lse.data_ptr<float>(),
                           dlossf.data_ptr<float>(),
                           dlogits.data_ptr<float>(), B, C);
    }
    HIP_CHECK_LAST();
    return dlogits;
}

// ------------------------------------------------------- argmax + correct ---

// first-max-index argmax per row (torch CPU tie semantics), compare with
// target, deterministic integer block+atomic accumulation.
template <typename T>
__global__ void k_argmax_correct(const T* __restrict__ logits,
                                 const long* __restrict__ target,
                                 long long* __restrict__ out, int B, int C) {
    __shared__ long long scratch[CE_BLOCK / WAVE];
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    long long correct = 0;
    if (i < B) {
        const T* row = logits + (long)i * C;
        float best = (float)row[0];
        int arg = 0;
        for (int c = 1; c < C; ++c) {
            float v = (float)row[c];
            if (v > best) { best = v; arg = c; }
        }
        correct = (arg == (int)target[i]) ? 1 : 0;
    }
    long long total = lds_block_reduce_ll(correct, scratch);
    if (threadIdx.x == 0 && total)
        atomicAdd(reinterpret_cast<unsigned long long*>(out),
                  (unsigned long long)total);
}

at::Tensor argmax_correct(at::Tensor logits, at::Tensor target) {
    TORCH_CHECK(logits.dim() == 2 && logits.is_cuda());
    auto lc = logits.contiguous();
    int B = lc.size(0), C = lc.size(1);
    auto out = at::zeros({}, lc.options().dtype(at::kLong));
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = ceil_div_i(B, CE_BLOCK);
    if (lc.scalar_type() == at::kBFloat16) {
        hipLaunchKernelGGL(k_argmax_correct<bf16>, dim3(blocks), dim3(CE_BLOCK),
                           0, stream.stream(),
                           reinterpret_cast<const bf16*>(lc.data_ptr()),
                           target.data_ptr<long>(),
                           reinterpret_cast<long long*>(out.data_ptr()), B, C);
    } else {
        hipLaunchKernelGGL(k_argmax_correct<float>, dim3(blocks), dim3(CE_BLOCK),
                           0, stream.stream(), lc.data_ptr<float>(),
                           target.data_ptr<long>(),
                           reinterpret_cast<long long*>(out.data_ptr()), B, C);
    }
    HIP_CHECK_LAST();
    return out;
}
