// pool.hip — 2×2/stride-2 max pooling, NHWC (channels_last), bf16.
//
// SURVEY N9.  Forward stores a 2-bit argmax index per output element (u8);
// backward scatters dy into the window — windows are disjoint (stride 2)
// so every input element is written exactly once (no zero-init, no
// atomics, deterministic).  Lanes run along C (fastest dim) → coalesced.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

// bf16x8-vectorized: each thread owns an 8-channel group of one output
// pixel — 4×16 B window loads, 16 B result store, 8 B packed argmax.
__global__ void k_maxpool2x2_fwd(const bf16* __restrict__ x,
                                 bf16* __restrict__ y,
                                 unsigned char* __restrict__ idx,
                                 long total_v, int Ho, int Wo, int Cv,
                                 int H, int W) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    const int C = Cv * 8;
    for (; i < total_v; i += stride) {
        int c8 = i % Cv;
        long r = i / Cv;
        int wo = r % Wo; r /= Wo;
        int ho = r % Ho; long n = r / Ho;
        const s16x8* base = reinterpret_cast<const s16x8*>(
            x + ((n * H + 2 * ho) * W + 2 * wo) * C) + c8;
        const int wv = W * Cv;
        s16x8 v0 = base[0];
        s16x8 v1 = base[Cv];
        s16x8 v2 = base[wv];
        s16x8 v3 = base[wv + Cv];
        s16x8 o;
        unsigned long long packed = 0;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            float b = us2f((unsigned short)v0[j]);
            int arg = 0;
            float f1 = us2f((unsigned short)v1[j]);
            if (f1 > b) { b = f1; arg = 1; }
            float f2 = us2f((unsigned short)v2[j]);
            if (f2 > b) { b = f2; arg = 2; }
            float f3 = us2f((unsigned short)v3[j]);
            if (f3 > b) { b = f3; arg = 3; }
            o[j] = (short)f2us(b);
            packed |= (unsigned long long)arg << (8 * j);
        }
        reinterpret_cast<s16x8*>(y)[i] = o;
        reinterpret_cast<unsigned long long*>(idx)[i] = packed;
    }
}

__global__ void k_maxpool2x2_bwd(const bf16* __restrict__ dy,
                                 const unsigned char* __restrict__ idx,
                                 bf16* __restrict__ dx,
                                 long total_v, int Ho, int Wo, int Cv,
                                 int H, int W) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    const int C = Cv * 8;
    for (; i < total_v; i += stride) {
        int c8 = i % Cv;
        long r = i / Cv;
        int wo = r % Wo; r /= Wo;
        int ho = r % Ho; long n = r / Ho;
        s16x8* base = reinterpret_cast<s16x8*>(
            dx + ((n * H + 2 * ho) * W + 2 * wo) * C) + c8;
        const int wv = W * Cv;
        s16x8 g = reinterpret_cast<const s16x8*>(dy)[i];
        unsigned long long packed =
            reinterpret_cast<const unsigned long long*>(idx)[i];
        s16x8 o0 = {}, o1 = {}, o2 = {}, o3 = {};
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            int arg = (packed >> (8 * j)) & 0xff;
            short gv = g[j];
            if (arg == 0) o0[j] = gv;
            else if (arg == 1) o1[j] = gv;
            else if (arg == 2) o2[j] = gv;
            else o3[j] = gv;
        }
        base[0] = o0;
        base[Cv] = o1;
        base[wv] = o2;
        base[wv + Cv] = o3;
    }
}

// General max pooling (kernel ks, stride st, padding pad) for the ResNet
// stem (3×3/2/1) and friends.  Forward stores the flat window argmax
// (< ks*ks <= 256) per output element; backward GATHERS: windows may
// overlap (st < ks), so each input element sums the dy of every output
// window that claimed it — no atomics, deterministic.
// bf16x8-vectorized: thread owns an 8-channel group of one output pixel;
// argmax bytes packed 8-at-a-time (same layout trick as the 2x2 kernel).
__global__ void k_maxpool_fwd(const bf16* __restrict__ x,
                              bf16* __restrict__ y,
                              unsigned char* __restrict__ idx, long total_v,
                              int Ho, int Wo, int Cv, int H, int W,
                              int ks, int st, int pad) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    const int C = Cv * 8;
    for (; i < total_v; i += stride) {
        int c8 = i % Cv;
        long r = i / Cv;
        int wo = r % Wo; r /= Wo;
        int ho = r % Ho; long n = r / Ho;
        float best[8];
        int arg[8];
        #pragma unroll
        for (int j = 0; j < 8; ++j) { best[j] = -3.4e38f; arg[j] = 0; }
        for (int kh = 0; kh < ks; ++kh) {
            int h = ho * st + kh - pad;
            if (h < 0 || h >= H) continue;
            for (int kw = 0; kw < ks; ++kw) {
                int w = wo * st + kw - pad;
                if (w < 0 || w >= W) continue;
                s16x8 v = reinterpret_cast<const s16x8*>(
                    x + ((n * H + h) * W + w) * C)[c8];
                #pragma unroll
                for (int j = 0; j < 8; ++j) {
                    float f = us2f((unsigned short)v[j]);
                    if (f > best[j]) { best[j] = f; arg[j] = kh * ks + kw; }
                }
            }
        }
        s16x8 o;
        unsigned long long packed = 0;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            o[j] = (short)f2us(best[j]);
            packed |= (unsigned long long)(unsigned)arg[j] << (8 * j);
        }
        reinterpret_cast<s16x8*>(y)[i] = o;
        reinterpret_cast<unsigned long long*>(idx)[i] = packed;
    }
}

__global__ void k_maxpool_bwd(const bf16* __restrict__ dy,
                              const unsigned char* __restrict__ idx,
                              bf16* __restrict__ dx, long total_v,
                              int Ho, int Wo, int Cv, int H, int W,
                              int ks, int st, int pad) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    const int C = Cv * 8;
    for (; i < total_v; i += stride) {
        int c8 = i % Cv;
        long r = i / Cv;
        int w = r % W; r /= W;
        int h = r % H; long n = r / H;
        float acc[8] = {};
        // output windows containing (h, w): ho*st <= h+pad <= ho*st+ks-1
        int ho_lo = (h + pad - ks + 1 + st - 1); ho_lo = ho_lo > 0 ? ho_lo / st : 0;
        int ho_hi = min((h + pad) / st, Ho - 1);
        int wo_lo = (w + pad - ks + 1 + st - 1); wo_lo = wo_lo > 0 ? wo_lo / st : 0;
        int wo_hi = min((w + pad) / st, Wo - 1);
        for (int ho = ho_lo; ho <= ho_hi; ++ho) {
            int kh = h + pad - ho * st;
            for (int wo = wo_lo; wo <= wo_hi; ++wo) {
                int kw = w + pad - wo * st;
                long o = ((n * Ho + ho) * Wo + wo) * Cv + c8;
                unsigned char want = (unsigned char)(kh * ks + kw);
                unsigned long long packed =
                    reinterpret_cast<const unsigned long long*>(idx)[o];
                s16x8 g = reinterpret_cast<const s16x8*>(dy)[o];
                #pragma unroll
                for (int j = 0; j < 8; ++j)
                    if (((packed >> (8 * j)) & 0xff) == want)
                        acc[j] += us2f((unsigned short)g[j]);
            }
        }
        s16x8 o8;
        #pragma unroll
        for (int j = 0; j < 8; ++j) o8[j] = (short)f2us(acc[j]);
        reinterpret_cast<s16x8*>(dx)[i] = o8;
    }
}

// Global average pool [N,C,H,W] (channels_last) -> [N,C] fp32-accumulated.
// One block per n, lanes own channel groups, walkers sweep H*W rows.
__global__ void k_gap_fwd(const bf16* __restrict__ x, bf16* __restrict__ y,
                          int HW, int C) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* red = reinterpret_cast<float*>(smem);    // [nw][C]
    const int cpg = C >> 3;
    const int c8 = threadIdx.x % cpg;
    const int walker = threadIdx.x / cpg;
    const int nw = blockDim.x / cpg;
    const long n = blockIdx.x;
    float sum[8] = {};
    for (int r = walker; r < HW; r += nw) {
        s16x8 v = reinterpret_cast<const s16x8*>(x + (n * HW + r) * C)[c8];
        #pragma unroll
        for (int j = 0; j < 8; ++j) sum[j] += us2f((unsigned short)v[j]);
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) red[walker * C + c8 * 8 + j] = sum[j];
    __syncthreads();
    float inv = 1.f / HW;
    for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float t = 0.f;
        for (int w = 0; w < nw; ++w) t += red[w * C + c];
        y[n * C + c] = f2bf(t * inv);
    }
}

__global__ void k_gap_bwd(const bf16* __restrict__ dy, bf16* __restrict__ dx,
                          long total, int HW, int C) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    float inv = 1.f / HW;
    for (; i < total; i += stride) {
        int c = i % C;
        long n = i / C / HW;
        dx[i] = f2bf(bf2f(dy[n * C + c]) * inv);
    }
}

std::vector<at::Tensor> maxpool_fwd(at::Tensor x, long ks, long st, long pad) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    int Ho = (H + 2 * (int)pad - (int)ks) / (int)st + 1;
    int Wo = (W + 2 * (int)pad - (int)ks) / (int)st + 1;
    TORCH_CHECK(ks * ks <= 256);
    TORCH_CHECK(C % 8 == 0, "maxpool HIP path needs C % 8 == 0");
    auto y = at::empty({N, C, Ho, Wo},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
    auto idx = at::empty({N, Ho, Wo, C}, x.options().dtype(at::kByte));
    long total_v = (long)N * Ho * Wo * (C / 8);
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = std::min<long>(8192, ceil_div_i(total_v, 256));
    hipLaunchKernelGGL(k_maxpool_fwd, dim3(blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()),
                       idx.data_ptr<unsigned char>(), total_v, Ho, Wo, C / 8,
                       H, W, (int)ks, (int)st, (int)pad);
    HIP_CHECK_LAST();
    return {y, idx};
}

at::Tensor maxpool_bwd(at::Tensor dy, at::Tensor idx, long H, long W,
                       long ks, long st, long pad) {
    int N = dy.size(0), C = dy.size(1), Ho = dy.size(2), Wo = dy.size(3);
    auto dx = at::empty({N, C, (int)H, (int)W},
                        dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    long total_v = (long)N * H * W * (C / 8);
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = std::min<long>(8192, ceil_div_i(total_v, 256));
    hipLaunchKernelGGL(k_maxpool_bwd, dim3(blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(dy.data_ptr()),
                       idx.data_ptr<unsigned char>(),
                       reinterpret_cast<bf16*>(dx.data_ptr()),
                       total_v, Ho, Wo, C / 8, (int)H, (int)W,
                       (int)ks, (int)st, (int)pad);
    HIP_CHECK_LAST();
    return dx;
}

at::Tensor global_avgpool_fwd(at::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
    int N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
    TORCH_CHECK(C % 8 == 0 && C / 8 <= 256);
    auto y = at::empty({N, C}, x.options());
    int cpg = C / 8;
    int block = (256 / cpg) * cpg;
    int nw = block / cpg;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(k_gap_fwd, dim3(N), dim3(block), nw * C * 4,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()), HW, C);
    HIP_CHECK_LAST();
    return y;
}

at::Tensor global_avgpool_bwd(at::Tensor dy, long H, long W) {
    int N = dy.size(0), C = dy.size(1);
    auto dx = at::empty({N, C, (int)H, (int)W},
                        dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    long total = (long)N * H * W * C;
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = std::min<long>(4096, ceil_div_i(total, 256));
    hipLaunchKernelGGL(k_gap_bwd, dim3(blocks), dim3(256), 0, stream.stream(),
                       reinterpret_cast<const bf16*>(dy.data_ptr()),
                       reinterpret_cast<bf16*>(dx.data_ptr()),
                       total, (int)(H * W), C);
    HIP_CHECK_LAST();
    return dx;
}

std::vector<at::Tensor> maxpool2x2_fwd(at::Tensor x) {
    // x: NCHW logical, channels_last physical
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    TORCH_CHECK(H % 2 == 0 && W % 2 == 0, "maxpool2x2 requires even H,W");
    TORCH_CHECK(C % 8 == 0, "maxpool2x2 HIP path needs C % 8 == 0");
    int Ho = H / 2, Wo = W / 2;
    auto y = at::empty({N, C, Ho, Wo},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
    auto idx = at::empty({N, Ho, Wo, C}, x.options().dtype(at::kByte));
    long total_v = (long)N * Ho * Wo * (C / 8);
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = std::min<long>(8192, ceil_div_i(total_v, 256));
    hipLaunchKernelGGL(k_maxpool2x2_fwd, dim3(blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()),
                       idx.data_ptr<unsigned char>(), total_v, Ho, Wo, C / 8,
                       H, W);
    HIP_CHECK_LAST();
    return {y, idx};
}

at::Tensor maxpool2x2_bwd(at::Tensor dy, at::Tensor idx, long H, long W) {
    int N = dy.size(0), C = dy.size(1), Ho = dy.size(2), Wo = dy.size(3);
    auto dx = at::empty({N, C, (int)H, (int)W},
                        dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    long total_v = (long)N * Ho * Wo * (C / 8);
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = std::min<long>(8192, ceil_div_i(total_v, 256));
    hipLaunchKernelGGL(k_maxpool2x2_bwd, dim3(blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(dy.data_ptr()),
                       idx.data_ptr<unsigned char>(),
                       reinterpret_cast<bf16*>(dx.data_ptr()),
                       total_v, Ho, Wo, C / 8, (int)H, (int)W);
    HIP_CHECK_LAST();
    return dx;
}
