#include "hip/hip_runtime.h"
// pool.hip — 2×2/stride-2 max pooling, NHWC (channels_last), bf16.
//
// SURVEY N9.  Forward stores a 2-bit argmax index per output element (u8);
// backward scatters dy into the window — windows are disjoint (stride 2)
// so every input element is written exactly once (no zero-init, no
// atomics, deterministic).  Lanes run along C (fastest dim) → coalesced.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

__global__ void k_maxpool2x2_fwd(const bf16* __restrict__ x,
                                 bf16* __restrict__ y,
                                 unsigned char* __restrict__ idx,
                                 long total, int Ho, int Wo, int C, int H, int W) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int c = i % C;
        long r = i / C;
        int wo = r % Wo; r /= Wo;
        int ho = r % Ho; long n = r / Ho;
        const bf16* base = x + ((n * H + 2 * ho) * W + 2 * wo) * C + c;
        float v0 = bf2f(base[0]);
        float v1 = bf2f(base[C]);
        float v2 = bf2f(base[(long)W * C]);
        float v3 = bf2f(base[(long)W * C + C]);
        float best = v0; int arg = 0;
        if (v1 > best) { best = v1; arg = 1; }
        if (v2 > best) { best = v2; arg = 2; }
        if (v3 > best) { best = v3; arg = 3; }
        y[i] = f2bf(best);
        idx[i] = (unsigned char)arg;
    }
}

__global__ void k_maxpool2x2_bwd(const bf16* __restrict__ dy,
                                 const unsigned char* __restrict__ idx,
                                 bf16* __restrict__ dx,
                                 long total, int Ho, int Wo, int C, int H, int W) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    bf16 zero = f2bf(0.f);
    for (; i < total; i += stride) {
        int c = i % C;
        long r = i / C;
        int wo = r % Wo; r /= Wo;
        int ho = r % Ho; long n = r / Ho;
        bf16* base = dx + ((n * H + 2 * ho) * W + 2 * wo) * C + c;
        int arg = idx[i];
        bf16 g = dy[i];
        base[0] = (arg == 0) ? g : zero;
        base[C] = (arg == 1) ? g : zero;
        base[(long)W * C] = (arg == 2) ? g : zero;
        base[(long)W * C + C] = (arg == 3) ? g : zero;
    }
}

std::vector<at::Tensor> maxpool2x2_fwd(at::Tensor x) {
    // x: NCHW logical, channels_last physical
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    TORCH_CHECK(H % 2 == 0 && W % 2 == 0, "maxpool2x2 requires even H,W");
    int Ho = H / 2, Wo = W / 2;
    auto y = at::empty({N, C, Ho, Wo},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
    auto idx = at::empty({N, Ho, Wo, C}, x.options().dtype(at::kByte));
    long total = (long)N * Ho * Wo * C;
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = std::min<long>(4096, ceil_div_i(total, 256));
    hipLaunchKernelGGL(k_maxpool2x2_fwd, dim3(blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()),
                       idx.data_ptr<unsigned char>(), total, Ho, Wo, C, H, W);
    HIP_CHECK_LAST();
    return {y, idx};
}

at::Tensor maxpool2x2_bwd(at::Tensor dy, at::Tensor idx, long H, long W) {
    int N = dy.size(0), C = dy.size(1), Ho = dy.size(2), Wo = dy.size(3);
    auto dx = at::empty({N, C, (int)H, (int)W},
                        dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    long total = (long)N * Ho * Wo * C;
    auto stream = at::hip::getCurrentHIPStream();
    int blocks = std::min<long>(4096, ceil_div_i(total, 256));
    hipLaunchKernelGGL(k_maxpool2x2_bwd, dim3(blocks), dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(dy.data_ptr()),
                       idx.data_ptr<unsigned char>(),
                       reinterpret_cast<bf16*>(dx.data_ptr()),
                       total, Ho, Wo, C, (int)H, (int)W);
    HIP_CHECK_LAST();
    return dx;
}
