// Probe ds_read_b64_tr_b16 semantics on gfx950: fill LDS with element index,
// issue the transpose read with various per-lane address patterns, dump what
// each lane received.  (The programming guide's two descriptions disagree;
// wrong-data modes are silent, so measure.)
#include <hip/hip_runtime.h>
#include <cstdio>

typedef unsigned int u32;

__global__ void probe(unsigned short* out, int mode) {
    __shared__ unsigned short lds[1024];
    int l = threadIdx.x;
    for (int i = l; i < 1024; i += 64) lds[i] = (unsigned short)i;
    __syncthreads();
    unsigned short* addr;
    if (mode == 0)       addr = &lds[0];                          // uniform
    else if (mode == 1)  addr = &lds[(l & 15) * 4];               // T10: own 8B chunk
    else if (mode == 2)  addr = &lds[(l & 15) + (l >> 4) * 64];   // pattern base
    else                 addr = &lds[l * 4];                      // linear 8B
    typedef __attribute__((address_space(3))) unsigned short* lds_ptr;
    lds_ptr lp = (lds_ptr)addr;
    unsigned long long r;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(r)
                 : "v"(lp)
                 : "memory");
    out[(long)mode * 256 + l * 4 + 0] = (unsigned short)(r & 0xffff);
    out[(long)mode * 256 + l * 4 + 1] = (unsigned short)((r >> 16) & 0xffff);
    out[(long)mode * 256 + l * 4 + 2] = (unsigned short)((r >> 32) & 0xffff);
    out[(long)mode * 256 + l * 4 + 3] = (unsigned short)((r >> 48) & 0xffff);
}

int main() {
    unsigned short* d;
    hipMalloc(&d, 4 * 256 * sizeof(unsigned short));
    for (int mode = 0; mode < 4; ++mode)
        hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
    unsigned short h[4 * 256];
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    hipError_t e = hipGetLastError();
    if (e != hipSuccess) { printf("ERR %s\n", hipGetErrorString(e)); return 1; }
    for (int mode = 0; mode < 4; ++mode) {
        printf("=== mode %d ===\n", mode);
        for (int l = 0; l < 20; ++l)
            printf("lane %2d: %4d %4d %4d %4d\n", l,
                   h[mode * 256 + l * 4], h[mode * 256 + l * 4 + 1],
                   h[mode * 256 + l * 4 + 2], h[mode * 256 + l * 4 + 3]);
        printf("lane 16: %4d %4d %4d %4d\n",
               h[mode * 256 + 64], h[mode * 256 + 65], h[mode * 256 + 66],
               h[mode * 256 + 67]);
    }
    return 0;
}
