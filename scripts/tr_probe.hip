// Empirical mapping probe for gfx950 ds_read_b64_tr_b16.
// Fills LDS with identity u16 values, reads with the transpose-read at a
// configurable per-lane address pattern, and dumps lane -> 4 delivered
// element ids. From the dump we derive the exact gather so the attention
// V image can be laid out for it (guide T10).
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) short s16x4;

// pattern p: lane address (in elements; 4-element = 8B granularity)
__host__ __device__ int addr_for(int pat, int lane) {
    switch (pat) {
        case 0: return lane * 4;                       // contiguous granules
        case 1: return (lane & 15) * 4 + (lane >> 4) * 64;   // 16-lane groups
        case 2: return (lane & 3) * 16 + (lane >> 2) * 4;    // row-major 4x(16)
        case 3: return (lane & 3) * 64 + ((lane >> 2) & 3) * 4 + (lane >> 4) * 16;
        default: return lane * 4;
    }
}

__global__ void probe(short* out, int pat) {
    __shared__ unsigned short lds[2048];
    for (int i = threadIdx.x; i < 2048; i += blockDim.x) lds[i] = i;
    __syncthreads();
    const int lane = threadIdx.x & 63;
    s16x4 r = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (__attribute__((address_space(3))) s16x4*)&lds[addr_for(pat, lane)]);
    for (int j = 0; j < 4; ++j) out[lane * 4 + j] = r[j];
}

int main() {
    short* d;
    hipMalloc(&d, 64 * 4 * sizeof(short));
    short h[256];
    for (int pat = 0; pat < 4; ++pat) {
        hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, pat);
        hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
        printf("pattern %d (addr = lane-dependent, see addr_for):\n", pat);
        for (int l = 0; l < 64; ++l) {
            printf("  lane %2d addr %4d -> %4d %4d %4d %4d\n", l,
                   addr_for(pat, l),
                   h[l * 4], h[l * 4 + 1], h[l * 4 + 2], h[l * 4 + 3]);
        }
    }
    hipFree(d);
    return 0;
}
