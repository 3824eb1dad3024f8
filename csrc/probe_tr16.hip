// Standalone probe: empirically map ds_read_b64_tr_b16's lane<->LDS layout
// on gfx950 (no ISA doc in this environment).  Fills LDS with the identity
// pattern lds16[i] = i, issues the transpose-read at controlled per-lane
// addresses, dumps each lane's 4 delivered 16-bit values.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef short s16x4 __attribute__((ext_vector_type(4)));

__global__ void probe(short *out, int mode) {
    __shared__ __align__(16) short lds16[2048];
    int t = threadIdx.x;
    for (int i = t; i < 2048; i += 64) lds16[i] = (short)i;
    __syncthreads();
    unsigned base = (unsigned)(unsigned long long)(
        __attribute__((address_space(3))) short *)lds16;
    unsigned addr;
    if (mode == 0) addr = base + t * 8;              // linear 8B per lane
    else if (mode == 1) addr = base;                 // uniform
    else addr = base + ((t & 15) * 2 + (t >> 4) * 128);  // guide-ish guess
    s16x4 v;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=&v"(v)
                 : "v"(addr)
                 : "memory");
    for (int j = 0; j < 4; ++j) out[t * 4 + j] = v[j];
}

int main() {
    short *d;
    hipMalloc(&d, 64 * 4 * sizeof(short));
    short h[256];
    for (int mode = 0; mode < 3; ++mode) {
        hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
        hipDeviceSynchronize();
        hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
        printf("mode %d:\n", mode);
        for (int l = 0; l < 64; ++l) {
            printf("  lane %2d: %4d %4d %4d %4d\n", l, h[l * 4], h[l * 4 + 1],
                   h[l * 4 + 2], h[l * 4 + 3]);
        }
    }
    hipFree(d);
    return 0;
}
