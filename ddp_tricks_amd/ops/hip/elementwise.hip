// elementwise.hip — multi-tensor optimizer/amp kernels + casts.
//
// Implements the implicit native surface of apex amp_C + the fused SGD step
// (SURVEY N4/N12/N13): multi-tensor unscale+inf-check over the DDP bucket
// flats, fused SGD(momentum, nesterov) [+ Lookahead interpolation] over the
// whole parameter list in one launch, bf16<->fp32 casts.
//
// Multi-tensor scheme: host packs up to MT_MAX tensor pointers + a prefix
// of chunk offsets into the kernarg struct; blocks map to (tensor, chunk)
// so one launch covers the whole list.  16 Ki-element chunks give ≳350
// blocks on the 23 MB flagship parameter set (256-CU chip wants ≫256
// workgroups); bodies are f32x4-vectorized.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

#include <vector>

constexpr int MT_MAX = 32;
constexpr int MT_CHUNK = 1 << 14;  // elements per block-chunk
constexpr int MT_BLOCK = 256;

struct MTMeta {
    const float* __restrict__ a[MT_MAX];
    float* __restrict__ b[MT_MAX];
    float* __restrict__ c[MT_MAX];
    long sizes[MT_MAX];
    int chunk_start[MT_MAX + 1];  // prefix sum of per-tensor chunk counts
    int ntensors;
};

DEV_INLINE int find_tensor(const MTMeta& m, int chunk, int& local_chunk) {
    int t = 0;
    while (t + 1 < m.ntensors && m.chunk_start[t + 1] <= chunk) ++t;
    local_chunk = chunk - m.chunk_start[t];
    return t;
}

DEV_INLINE void chunk_range(const MTMeta& m, int& t, long& base, long& end) {
    int local;
    t = find_tensor(m, blockIdx.x, local);
    long n = m.sizes[t];
    base = (long)local * MT_CHUNK;
    end = base + MT_CHUNK < n ? base + MT_CHUNK : n;
}

// ---------------------------------------------------------------- unscale ---

__global__ void k_mt_unscale(MTMeta m, float inv_scale, float* found_inf) {
    int t; long base, end;
    chunk_range(m, t, base, end);
    float* g = m.b[t];
    long vb = base >> 2, ve = end >> 2;
    bool bad = false;
    for (long i = vb + threadIdx.x; i < ve; i += MT_BLOCK) {
        f32x4 v = reinterpret_cast<f32x4*>(g)[i];
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            v[j] *= inv_scale;
            if (!isfinite(v[j])) bad = true;
        }
        reinterpret_cast<f32x4*>(g)[i] = v;
    }
    for (long i = ve * 4 + threadIdx.x; i < end; i += MT_BLOCK) {
        float v = g[i] * inv_scale;
        if (!isfinite(v)) bad = true;
        g[i] = v;
    }
    if (bad) *found_inf = 1.0f;  // racy same-value store: any writer sets it
}

void multi_tensor_unscale(std::vector<at::Tensor> grads, at::Tensor found_inf,
                          double inv_scale) {
    auto stream = at::hip::getCurrentHIPStream();
    for (size_t start = 0; start < grads.size(); start += MT_MAX) {
        MTMeta m{};
        int chunks = 0;
        int nt = 0;
        for (size_t i = start; i < std::min(grads.size(), start + MT_MAX); ++i) {
            auto& g = grads[i];
            TORCH_CHECK(g.is_cuda() && g.scalar_type() == at::kFloat &&
                        g.is_contiguous(), "unscale expects contiguous fp32");
            m.b[nt] = g.data_ptr<float>();
            m.sizes[nt] = g.numel();
            m.chunk_start[nt] = chunks;
            chunks += ceil_div_i(g.numel(), MT_CHUNK);
            ++nt;
        }
        m.chunk_start[nt] = chunks;
        m.ntensors = nt;
        if (chunks == 0) continue;
        hipLaunchKernelGGL(k_mt_unscale, dim3(chunks), dim3(MT_BLOCK), 0,
                           stream.stream(), m, (float)inv_scale,
                           found_inf.data_ptr<float>());
        HIP_CHECK_LAST();
    }
}

// -------------------------------------------------------------- fused SGD ---

// buf = mu*buf + g ; d = nesterov ? g + mu*buf : buf ; p -= lr*d
// With found_inf set, the step is a device-side no-op (amp overflow skip
// without a host sync).
__global__ void k_mt_sgd(MTMeta m, float lr, float mu, float wd, float nesterov,
                         const float* __restrict__ found_inf) {
    if (found_inf && *found_inf != 0.0f) return;
    int t; long base, end;
    chunk_range(m, t, base, end);
    const float* __restrict__ g = m.a[t];
    float* __restrict__ p = m.b[t];
    float* __restrict__ buf = m.c[t];
    long vb = base >> 2, ve = end >> 2;
    for (long i = vb + threadIdx.x; i < ve; i += MT_BLOCK) {
        f32x4 gv = reinterpret_cast<const f32x4*>(g)[i];
        f32x4 pv = reinterpret_cast<f32x4*>(p)[i];
        f32x4 bv = mu != 0.f ? reinterpret_cast<f32x4*>(buf)[i] : f32x4{};
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            float gi = gv[j];
            if (wd != 0.0f) gi = fmaf(wd, pv[j], gi);
            float d = gi;
            if (mu != 0.0f) {
                float b = fmaf(mu, bv[j], gi);
                bv[j] = b;
                d = (nesterov != 0.0f) ? fmaf(mu, b, gi) : b;
            }
            pv[j] = fmaf(-lr, d, pv[j]);
        }
        reinterpret_cast<f32x4*>(p)[i] = pv;
        if (mu != 0.f) reinterpret_cast<f32x4*>(buf)[i] = bv;
    }
    for (long i = ve * 4 + threadIdx.x; i < end; i += MT_BLOCK) {
        float gi = g[i];
        if (wd != 0.0f) gi = fmaf(wd, p[i], gi);
        float d = gi;
        if (mu != 0.0f) {
            float b = fmaf(mu, buf[i], gi);
            buf[i] = b;
            d = (nesterov != 0.0f) ? fmaf(mu, b, gi) : b;
        }
        p[i] = fmaf(-lr, d, p[i]);
    }
}

void fused_sgd(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
               std::vector<at::Tensor> bufs, double lr, double momentum,
               double wd, double nesterov,
               c10::optional<at::Tensor> found_inf) {
    TORCH_CHECK(params.size() == grads.size());
    bool has_mu = momentum != 0.0;
    TORCH_CHECK(!has_mu || bufs.size() == params.size());
    auto stream = at::hip::getCurrentHIPStream();
    const float* fi = found_inf.has_value()
        ? found_inf->data_ptr<float>() : nullptr;
    for (size_t start = 0; start < params.size(); start += MT_MAX) {
        MTMeta m{};
        int chunks = 0, nt = 0;
        for (size_t i = start; i < std::min(params.size(), start + MT_MAX); ++i) {
            m.a[nt] = grads[i].data_ptr<float>();
            m.b[nt] = params[i].data_ptr<float>();
            m.c[nt] = has_mu ? bufs[i].data_ptr<float>() : nullptr;
            m.sizes[nt] = params[i].numel();
            m.chunk_start[nt] = chunks;
            chunks += ceil_div_i(params[i].numel(), MT_CHUNK);
            ++nt;
        }
        m.chunk_start[nt] = chunks;
        m.ntensors = nt;
        if (chunks == 0) continue;
        hipLaunchKernelGGL(k_mt_sgd, dim3(chunks), dim3(MT_BLOCK), 0,
                           stream.stream(), m, (float)lr, (float)momentum,
                           (float)wd, (float)nesterov, fi);
        HIP_CHECK_LAST();
    }
}

// --------------------------------------------------------------- lookahead ---

// slow += alpha * (fast - slow); fast = slow     (reference lookahead.py:19-27)
__global__ void k_mt_lookahead(MTMeta m, float alpha,
                               const float* __restrict__ found_inf) {
    if (found_inf && *found_inf != 0.0f) return;
    int t; long base, end;
    chunk_range(m, t, base, end);
    float* __restrict__ fast = m.b[t];
    float* __restrict__ slow = m.c[t];
    long vb = base >> 2, ve = end >> 2;
    for (long i = vb + threadIdx.x; i < ve; i += MT_BLOCK) {
        f32x4 fv = reinterpret_cast<f32x4*>(fast)[i];
        f32x4 sv = reinterpret_cast<f32x4*>(slow)[i];
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            float s = fmaf(alpha, fv[j] - sv[j], sv[j]);
            sv[j] = s;
            fv[j] = s;
        }
        reinterpret_cast<f32x4*>(fast)[i] = fv;
        reinterpret_cast<f32x4*>(slow)[i] = sv;
    }
    for (long i = ve * 4 + threadIdx.x; i < end; i += MT_BLOCK) {
        float s = fmaf(alpha, fast[i] - slow[i], slow[i]);
        slow[i] = s;
        fast[i] = s;
    }
}

void fused_lookahead(std::vector<at::Tensor> fast, std::vector<at::Tensor> slow,
                     double alpha, c10::optional<at::Tensor> found_inf) {
    TORCH_CHECK(fast.size() == slow.size());
    auto stream = at::hip::getCurrentHIPStream();
    const float* fi = found_inf.has_value()
        ? found_inf->data_ptr<float>() : nullptr;
    for (size_t start = 0; start < fast.size(); start += MT_MAX) {
        MTMeta m{};
        int chunks = 0, nt = 0;
        for (size_t i = start; i < std::min(fast.size(), start + MT_MAX); ++i) {
            m.b[nt] = fast[i].data_ptr<float>();
            m.c[nt] = slow[i].data_ptr<float>();
            m.sizes[nt] = fast[i].numel();
            m.chunk_start[nt] = chunks;
            chunks += ceil_div_i(fast[i].numel(), MT_CHUNK);
            ++nt;
        }
        m.chunk_start[nt] = chunks;
        m.ntensors = nt;
        if (chunks == 0) continue;
        hipLaunchKernelGGL(k_mt_lookahead, dim3(chunks), dim3(MT_BLOCK), 0,
                           stream.stream(), m, (float)alpha, fi);
        HIP_CHECK_LAST();
    }
}

// ------------------------------------------------------------------- casts ---

__global__ void k_f32_to_bf16(const float* __restrict__ in,
                              unsigned short* __restrict__ out, long n) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    long nv = n >> 2;
    for (long v = i; v < nv; v += stride) {
        f32x4 f = reinterpret_cast<const f32x4*>(in)[v];
        s16x4 o;
        o[0] = f2us(f[0]); o[1] = f2us(f[1]); o[2] = f2us(f[2]); o[3] = f2us(f[3]);
        reinterpret_cast<s16x4*>(out)[v] = o;
    }
    for (long j = nv * 4 + i; j < n; j += stride)
        out[j] = f2us(in[j]);
}

at::Tensor cast_to_bf16(at::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kFloat && x.is_contiguous());
    auto y = at::empty_like(x, x.options().dtype(at::kBFloat16));
    long n = x.numel();
    int blocks = (int)std::min<long>(2048, (n / 4 + 255) / 256 + 1);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(k_f32_to_bf16, dim3(blocks), dim3(256), 0, stream.stream(),
                       x.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(y.data_ptr()), n);
    HIP_CHECK_LAST();
    return y;
}

// ------------------------------------------------- conv weight packing ---

// One pass over the fp32 master weight [K,C,R,S] producing BOTH bf16
// operand layouts: nhwc [K,R,S,Cp] (forward) and wt2 [Cp, R*S*K] (dgrad),
// with optional zero-padding of C to Cp=8 (the C<8 stem path).  Replaces
// the per-step chain of to(bf16) + contiguous(channels_last) + permute +
// reshape + contiguous ATen launches (SURVEY N4's cast layer).
__global__ void k_pack_conv_weight(const float* __restrict__ w,
                                   bf16* __restrict__ nhwc,
                                   bf16* __restrict__ wt2,
                                   int K, int C, int RS, int Cp) {
    long total = (long)K * Cp * RS;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int c = i % Cp;
        long r2 = i / Cp;
        int rs = r2 % RS;
        int k = r2 / RS;
        float v = c < C ? w[((long)k * C + c) * RS + rs] : 0.f;
        bf16 b = f2bf(v);
        nhwc[((long)k * RS + rs) * Cp + c] = b;           // [K,R,S,Cp]
        if (wt2)
            wt2[((long)c * RS + rs) * K + k] = b;         // [Cp, RS*K]
    }
}

std::vector<at::Tensor> pack_conv_weight(at::Tensor w, bool pad8,
                                         bool want_wt2) {
    TORCH_CHECK(w.is_cuda() && w.scalar_type() == at::kFloat &&
                w.is_contiguous());
    int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
    int Cp = pad8 ? 8 : C;
    int RS = R * S;
    auto opts = w.options().dtype(at::kBFloat16);
    // nhwc tensor carries channels_last metadata for [K,Cp,R,S]
    auto nhwc = at::empty({K, Cp, R, S},
                          opts.memory_format(at::MemoryFormat::ChannelsLast));
    at::Tensor wt2;
    bf16* wt2p = nullptr;
    if (want_wt2) {
        wt2 = at::empty({Cp, (long)RS * K}, opts);
        wt2p = reinterpret_cast<bf16*>(wt2.data_ptr());
    }
    long total = (long)K * Cp * RS;
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = std::min<long>(2048, ceil_div_i(total, 256));
    hipLaunchKernelGGL(k_pack_conv_weight, dim3(blocks), dim3(256), 0,
                       stream.stream(), w.data_ptr<float>(),
                       reinterpret_cast<bf16*>(nhwc.data_ptr()), wt2p,
                       K, C, RS, Cp);
    HIP_CHECK_LAST();
    if (want_wt2) return {nhwc, wt2};
    return {nhwc};
}

// Zero-pad channels to 8 in ONE pass (replaces at::zeros + slice copy_ —
// two ATen launches per step on the C<8 stem path).
__global__ void k_pad8(const bf16* __restrict__ x, bf16* __restrict__ xp,
                       long total_px, int C) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total_px; i += stride) {
        s16x8 o = {};
        const bf16* src = x + i * C;
        for (int c = 0; c < C; ++c) o[c] = *(const short*)&src[c];
        reinterpret_cast<s16x8*>(xp)[i] = o;
    }
}

at::Tensor pad8_channels(at::Tensor x) {
    // x: [N,C<8,H,W] channels_last bf16 -> [N,8,H,W] channels_last
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    TORCH_CHECK(C < 8);
    auto xp = at::empty({N, 8, H, W},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
    long total = (long)N * H * W;
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = std::min<long>(4096, ceil_div_i(total, 256));
    hipLaunchKernelGGL(k_pad8, dim3(blocks), dim3(256), 0, stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(xp.data_ptr()), total, C);
    HIP_CHECK_LAST();
    return xp;
}

// NHWC flatten boundary (Toy_Net conv->dense junction): [N,C,H,W]
// channels_last bf16 -> [N, C*H*W] in NCHW semantic order (the order the
// reference's fc1 weight layout expects; reference utils/model.py:23-25),
// plus the inverse for backward.  Replaces two ~23 us strided ATen
// permute-copies per step: here the fwd gather is coalesced across the
// c-threads (consecutive c -> consecutive addresses within each s row)
// and each thread writes its HW-contiguous chunk with s16x8 stores.
__global__ void k_nhwc_flatten(const bf16* __restrict__ x,
                               bf16* __restrict__ y, int C, int HW) {
    const int c = blockIdx.y * blockDim.x + threadIdx.x;
    const long b = blockIdx.x;
    if (c >= C) return;
    const bf16* src = x + (long)b * HW * C + c;
    bf16* dst = y + ((long)b * C + c) * HW;
    // NO runtime-indexed staging array (it would spill to scratch — the
    // HW bound is runtime); fixed-8 inner unroll keeps the gather in
    // registers and the store a single s16x8.
    if ((HW & 7) == 0) {
        for (int s8 = 0; s8 < HW; s8 += 8) {
            s16x8 v;
            #pragma unroll
            for (int k = 0; k < 8; ++k)
                v[k] = *(const short*)&src[(long)(s8 + k) * C];
            *reinterpret_cast<s16x8*>(&dst[s8]) = v;
        }
    } else {
        for (int s = 0; s < HW; ++s)
            *(short*)&dst[s] = *(const short*)&src[(long)s * C];
    }
}

__global__ void k_nhwc_unflatten(const bf16* __restrict__ dy,
                                 bf16* __restrict__ dx, int C, int HW) {
    const int c = blockIdx.y * blockDim.x + threadIdx.x;
    const long b = blockIdx.x;
    if (c >= C) return;
    const bf16* src = dy + ((long)b * C + c) * HW;   // contiguous chunk
    bf16* dst = dx + (long)b * HW * C + c;
    if ((HW & 7) == 0) {
        for (int s8 = 0; s8 < HW; s8 += 8) {
            const s16x8 v = *reinterpret_cast<const s16x8*>(&src[s8]);
            #pragma unroll
            for (int k = 0; k < 8; ++k)
                *(short*)&dst[(long)(s8 + k) * C] = v[k];
        }
    } else {
        for (int s = 0; s < HW; ++s)
            *(short*)&dst[(long)s * C] = *(const short*)&src[s];
    }
}

at::Tensor nhwc_flatten(at::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 &&
                x.dim() == 4, "nhwc_flatten: 4D cuda bf16 expected");
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
                "nhwc_flatten: channels_last expected");
    const int N = x.size(0), C = x.size(1);
    const int HW = x.size(2) * x.size(3);
    TORCH_CHECK(HW <= 64, "nhwc_flatten: HW too large");
    auto y = at::empty({(long)N, (long)C * HW}, x.options());
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(k_nhwc_flatten,
                       dim3(N, ceil_div_i(C, 256)), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()), C, HW);
    HIP_CHECK_LAST();
    return y;
}

at::Tensor nhwc_unflatten(at::Tensor dy, long C, long H, long W) {
    TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16 &&
                dy.dim() == 2 && dy.is_contiguous(),
                "nhwc_unflatten: 2D contiguous cuda bf16 expected");
    const long N = dy.size(0);
    const int HW = (int)(H * W);
    TORCH_CHECK(dy.size(1) == C * HW && HW <= 64);
    auto dx = at::empty({N, C, H, W},
                        dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(k_nhwc_unflatten,
                       dim3(N, ceil_div_i((int)C, 256)), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(dy.data_ptr()),
                       reinterpret_cast<bf16*>(dx.data_ptr()), (int)C, HW);
    HIP_CHECK_LAST();
    return dx;
}
