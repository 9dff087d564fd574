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
#include <climits>
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

// Wave-per-row variants for wide rows (the ResNet-50 [B,1000] head —
// the thread-per-row kernel walks 3×C serially,
// 254 µs for ~4 MB of reads).  One 64-lane wave owns a row; lanes stream
// 16 B bf16x8 chunks (coalesced 1 KiB per wave-iteration) with an online
// softmax (single pass over the row), then a shfl tree combines the
// per-lane (max, scaled-sum) pairs.  bf16 rows with C % 8 == 0 only;
// everything else keeps the scalar kernels.
constexpr int CE_WPB = CE_BLOCK / WAVE;   // rows (waves) per block

DEV_INLINE void ce_online8(const s16x8 v, float& m, float& s) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        float x = us2f((unsigned short)v[j]);
        if (x > m) {
            s *= expf(m - x);   // m == -inf ⇒ s == 0, exp(-inf)=0 handled
            m = x;
        }
        s += expf(x - m);
    }
}

__global__ void k_ce_rows_wave(const bf16* __restrict__ logits,
                               const long* __restrict__ target,
                               float* __restrict__ rowloss,
                               float* __restrict__ lse_out,
                               int B, int C) {
    const int row_i = blockIdx.x * CE_WPB + ((int)threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    if (row_i >= B) return;
    const bf16* row = logits + (long)row_i * C;
    const int chunks = C >> 3;
    float m = -INFINITY, s = 0.f;
    for (int ch = lane; ch < chunks; ch += WAVE) {
        s16x8 v = *reinterpret_cast<const s16x8*>(row + ch * 8);
        ce_online8(v, m, s);
    }
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        float mo = __shfl_down(m, off, 64);
        float so = __shfl_down(s, off, 64);
        float mn = fmaxf(m, mo);
        float t1 = (s == 0.f) ? 0.f : s * expf(m - mn);
        float t2 = (so == 0.f) ? 0.f : so * expf(mo - mn);
        m = mn;
        s = t1 + t2;
    }
    if (lane == 0) {
        float lse = logf(s) + m;
        lse_out[row_i] = lse;
        rowloss[row_i] = lse - (float)row[target[row_i]];
    }
}

__global__ void k_ce_bwd_wave(const bf16* __restrict__ logits,
                              const long* __restrict__ target,
                              const float* __restrict__ lse,
                              const float* __restrict__ dloss,
                              bf16* __restrict__ dlogits, int B, int C) {
    const int row_i = blockIdx.x * CE_WPB + ((int)threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    if (row_i >= B) return;
    const float g = dloss[0] / B;
    const bf16* row = logits + (long)row_i * C;
    bf16* drow = dlogits + (long)row_i * C;
    const float l = lse[row_i];
    const int t = (int)target[row_i];
    const int chunks = C >> 3;
    for (int ch = lane; ch < chunks; ch += WAVE) {
        s16x8 v = *reinterpret_cast<const s16x8*>(row + ch * 8);
        s16x8 o;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            int c = ch * 8 + j;
            float p = expf(us2f((unsigned short)v[j]) - l);
            o[j] = (short)f2us((p - (c == t ? 1.f : 0.f)) * g);
        }
        *reinterpret_cast<s16x8*>(drow + ch * 8) = o;
    }
}

__global__ void k_argmax_correct_wave(const bf16* __restrict__ logits,
                                      const long* __restrict__ target,
                                      long long* __restrict__ out,
                                      int B, int C) {
    const int row_i = blockIdx.x * CE_WPB + ((int)threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    if (row_i >= B) return;
    const bf16* row = logits + (long)row_i * C;
    const int chunks = C >> 3;
    float best = -INFINITY;
    int arg = INT_MAX;
    for (int ch = lane; ch < chunks; ch += WAVE) {
        s16x8 v = *reinterpret_cast<const s16x8*>(row + ch * 8);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float x = us2f((unsigned short)v[j]);
            if (x > best) { best = x; arg = ch * 8 + j; }
        }
    }
    // combine with torch's first-max tie rule: larger value wins, equal
    // value ⇒ smaller index wins
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        float bo = __shfl_down(best, off, 64);
        int ao = __shfl_down(arg, off, 64);
        if (bo > best || (bo == best && ao < arg)) { best = bo; arg = ao; }
    }
    if (lane == 0 && arg == (int)target[row_i])
        atomicAdd(reinterpret_cast<unsigned long long*>(out), 1ull);
}

static inline bool ce_wave_ok(const at::Tensor& logits, int C) {
    return logits.scalar_type() == at::kBFloat16 && (C % 8) == 0 && C >= 64;
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
    if (ce_wave_ok(logits, C)) {
        hipLaunchKernelGGL(k_ce_rows_wave, dim3(ceil_div_i(B, CE_WPB)),
                           dim3(CE_BLOCK), 0, stream.stream(),
                           reinterpret_cast<const bf16*>(logits.data_ptr()),
                           target.data_ptr<long>(), rowloss.data_ptr<float>(),
                           lse.data_ptr<float>(), B, C);
    } else if (logits.scalar_type() == at::kBFloat16) {
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
    if (ce_wave_ok(logits, C)) {
        hipLaunchKernelGGL(k_ce_bwd_wave, dim3(ceil_div_i(B, CE_WPB)),
                           dim3(CE_BLOCK), 0, stream.stream(),
                           reinterpret_cast<const bf16*>(logits.data_ptr()),
                           target.data_ptr<long>(), lse.data_ptr<float>(),
                           dlossf.data_ptr<float>(),
                           reinterpret_cast<bf16*>(dlogits.data_ptr()), B, C);
    } else if (logits.scalar_type() == at::kBFloat16) {
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
    if (ce_wave_ok(lc, C)) {
        hipLaunchKernelGGL(k_argmax_correct_wave, dim3(ceil_div_i(B, CE_WPB)),
                           dim3(CE_BLOCK), 0, stream.stream(),
                           reinterpret_cast<const bf16*>(lc.data_ptr()),
                           target.data_ptr<long>(),
                           reinterpret_cast<long long*>(out.data_ptr()), B, C);
    } else if (lc.scalar_type() == at::kBFloat16) {
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
