// Empirical operand-layout probe for gfx950's block-scaled MX MFMA
//   __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4 (fp8 e4m3 A and B)
// — the only large-K fp8 MFMA on gfx950 (guide §3: no non-scaled
// 32x32x64). C/D layout is shape-determined (identical to the bf16
// 32x32x16 map: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)); the
// A/B per-lane byte->(row,k) maps are undocumented here, so this probe
// packs A/B under CANDIDATE maps on the host and checks D = A@B against
// a host fp32 reference on exactly-representable fp8 values.
//
// Candidates (j = byte index 0..31, l32 = lane&31, hi = lane>>5):
//   0: A[row=l32][k = hi*32 + j]            (contiguous 32-chunk per half)
//   1: A[row=l32][k = hi*8 + (j&7) + 16*(j>>3)]   (8-chunks interleaved)
//   2: A[row=l32][k = j*2 + hi]             (element-interleaved halves)
//   3: A[row=l32][k = hi*16 + (j&15) + 32*(j>>4)] (16-chunks interleaved)
// B uses the same map with col=l32 (B is [k][col] fed per-lane like A).
// Also probes the scale operand: all-ones data, scaleA byte=128 (=2^1)
// must double D if scales apply; 127 = 2^0 identity.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 mx_probe.hip -o mx_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(8))) int i32x8;

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP error %d at %d\n", e, __LINE__); exit(1); } } while (0)

// host e4m3fn encode for exactly-representable values (halves in [-8, 8])
static unsigned char enc_e4m3(float v) {
    if (v == 0.f) return 0;
    unsigned char s = v < 0 ? 0x80 : 0;
    float a = fabsf(v);
    int e = (int)floorf(log2f(a));
    float m = a / exp2f((float)e) - 1.0f;  // in [0,1)
    int mant = (int)roundf(m * 8.0f);
    if (mant == 8) { mant = 0; e += 1; }
    int exp = e + 7;
    if (exp < 1 || exp > 15) { printf("enc range %f\n", v); exit(1); }
    return s | (unsigned char)(exp << 3) | (unsigned char)mant;
}

__global__ void mx_kernel(const int* a_img, const int* b_img, float* d_out,
                          int scale_a) {
    const int lane = threadIdx.x;
    i32x8 a, b;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
        a[i] = a_img[lane * 8 + i];
        b[i] = b_img[lane * 8 + i];
    }
    f32x16 c;
#pragma unroll
    for (int i = 0; i < 16; ++i) c[i] = 0.f;
    c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
        a, b, c, 0 /*cbsz: fp8*/, 0 /*blgp: fp8*/, 0, scale_a, 0, 127);
    const int col = lane & 31;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
        const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        d_out[row * 32 + col] = c[r];
    }
}

static int kmap(int cand, int hi, int j) {
    switch (cand) {
    case 0: return hi * 32 + j;
    case 1: return hi * 8 + (j & 7) + 16 * (j >> 3);
    case 2: return j * 2 + hi;
    default: return hi * 16 + (j & 15) + 32 * (j >> 4);
    }
}

int main() {
    // random exactly-representable values: {0, ±0.5, ±1, ±1.5, ±2}
    std::vector<float> A(32 * 64), B(64 * 32);
    unsigned x = 22222;
    auto rnd = [&]() {
        x = x * 1664525u + 1013904223u;
        return 0.5f * (float)((int)((x >> 20) % 9) - 4);
    };
    for (auto& v : A) v = rnd();
    for (auto& v : B) v = rnd();
    std::vector<float> ref(32 * 32, 0.f);
    for (int r = 0; r < 32; ++r)
        for (int c = 0; c < 32; ++c) {
            float acc = 0.f;
            for (int k = 0; k < 64; ++k) acc += A[r * 64 + k] * B[k * 32 + c];
            ref[r * 32 + c] = acc;
        }

    int *da, *db;
    float* dd;
    HIP_CHECK(hipMalloc(&da, 64 * 32));
    HIP_CHECK(hipMalloc(&db, 64 * 32));
    HIP_CHECK(hipMalloc(&dd, 32 * 32 * 4));
    std::vector<float> got(32 * 32);

    for (int cand = 0; cand < 4; ++cand) {
        std::vector<unsigned char> ha(64 * 32), hb(64 * 32);
        for (int lane = 0; lane < 64; ++lane) {
            const int l32 = lane & 31, hi = lane >> 5;
            for (int j = 0; j < 32; ++j) {
                const int k = kmap(cand, hi, j);
                ha[lane * 32 + j] = enc_e4m3(A[l32 * 64 + k]);
                hb[lane * 32 + j] = enc_e4m3(B[k * 32 + l32]);
            }
        }
        HIP_CHECK(hipMemcpy(da, ha.data(), 64 * 32, hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(db, hb.data(), 64 * 32, hipMemcpyHostToDevice));
        HIP_CHECK(hipMemset(dd, 0, 32 * 32 * 4));
        hipLaunchKernelGGL(mx_kernel, dim3(1), dim3(64), 0, 0, da, db, dd,
                           127);
        HIP_CHECK(hipGetLastError());
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipMemcpy(got.data(), dd, 32 * 32 * 4,
                            hipMemcpyDeviceToHost));
        int bad = 0;
        float maxd = 0.f;
        for (int i = 0; i < 32 * 32; ++i) {
            const float d = fabsf(got[i] - ref[i]);
            maxd = fmaxf(maxd, d);
            if (d > 1e-3f) ++bad;
        }
        printf("candidate %d: %s (bad=%d maxdiff=%g)\n", cand,
               bad == 0 ? "MATCH" : "no", bad, maxd);
        if (cand == 0) {
            // scale probe on whichever packing: byte 128 = 2^1 on A
            hipLaunchKernelGGL(mx_kernel, dim3(1), dim3(64), 0, 0, da, db,
                               dd, 128);
            HIP_CHECK(hipDeviceSynchronize());
            HIP_CHECK(hipMemcpy(got.data(), dd, 32 * 32 * 4,
                                hipMemcpyDeviceToHost));
            printf("scaleA=128 sample: got %g ref %g (x2 if scales live)\n",
                   got[5], ref[5]);
        }
    }
    (void)hipFree(da); (void)hipFree(db); (void)hipFree(dd);
    return 0;
}
