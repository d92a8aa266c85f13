// qk_norm_rope RPW (rows-per-wave) A/B (gfx950): the shipped kernel
// stages RPW=4 rows of q AND k before reducing (8 loads in flight);
// measured 3.45 TB/s in the flux step — test deeper batching.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 qknr_ab.hip -o qknr
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <algorithm>
#include <cmath>

using bf16 = __hip_bfloat16;
#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP error %d at %d\n", e, __LINE__); exit(1); } } while (0)
__device__ __forceinline__ float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2bf(float v) { return __float2bfloat16(v); }

template <int RPW>
__global__ void qknr_kernel(bf16* __restrict__ q, bf16* __restrict__ k,
                            const bf16* __restrict__ wq,
                            const bf16* __restrict__ wk,
                            const float* __restrict__ cs,
                            int S, int H, int D,
                            long q_bs, long q_hs, long q_ss,
                            long n_rows, float eps) {
    const int lane = threadIdx.x & 63;
    const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int pairs = D / 2;
    const bool act = lane < pairs;
    const int ln = act ? lane : 0;

    unsigned int uq[RPW], uk[RPW];
    unsigned int* qp[RPW];
    unsigned int* kp[RPW];
    float cvec[RPW], svec[RPW];
    const unsigned int uwq = reinterpret_cast<const unsigned int*>(wq)[ln];
    const unsigned int uwk = reinterpret_cast<const unsigned int*>(wk)[ln];
#pragma unroll
    for (int i = 0; i < RPW; ++i) {
        const long row = wave * RPW + i;
        const long rr = row < n_rows ? row : n_rows - 1;
        const long b = rr / ((long)S * H);
        const long sh = rr % ((long)S * H);
        const int sj = (int)(sh / H);
        const int h = (int)(sh % H);
        qp[i] = reinterpret_cast<unsigned int*>(
            q + b * q_bs + (long)sj * q_ss + (long)h * q_hs);
        kp[i] = reinterpret_cast<unsigned int*>(
            k + b * q_bs + (long)sj * q_ss + (long)h * q_hs);
        uq[i] = qp[i][ln];
        uk[i] = kp[i][ln];
        cvec[i] = cs[((long)sj * pairs + ln) * 2 + 0];
        svec[i] = cs[((long)sj * pairs + ln) * 2 + 1];
    }
    const float wq0 = bf2f(__ushort_as_bfloat16((unsigned short)(uwq & 0xffff)));
    const float wq1 = bf2f(__ushort_as_bfloat16((unsigned short)(uwq >> 16)));
    const float wk0 = bf2f(__ushort_as_bfloat16((unsigned short)(uwk & 0xffff)));
    const float wk1 = bf2f(__ushort_as_bfloat16((unsigned short)(uwk >> 16)));
#pragma unroll
    for (int i = 0; i < RPW; ++i) {
        if (wave * RPW + i >= n_rows) break;
        {
            float a0 = bf2f(__ushort_as_bfloat16((unsigned short)(uq[i] & 0xffff)));
            float a1 = bf2f(__ushort_as_bfloat16((unsigned short)(uq[i] >> 16)));
            float ss_ = act ? a0 * a0 + a1 * a1 : 0.f;
#pragma unroll
            for (int off = 32; off > 0; off >>= 1)
                ss_ += __shfl_xor(ss_, off, 64);
            const float rr = rsqrtf(ss_ / (float)D + eps);
            a0 = a0 * rr * wq0;
            a1 = a1 * rr * wq1;
            const float o0 = a0 * cvec[i] - a1 * svec[i];
            const float o1 = a0 * svec[i] + a1 * cvec[i];
            if (act)
                qp[i][lane] = (unsigned int)__bfloat16_as_ushort(f2bf(o0)) |
                              ((unsigned int)__bfloat16_as_ushort(f2bf(o1)) << 16);
        }
        {
            float a0 = bf2f(__ushort_as_bfloat16((unsigned short)(uk[i] & 0xffff)));
            float a1 = bf2f(__ushort_as_bfloat16((unsigned short)(uk[i] >> 16)));
            float ss_ = act ? a0 * a0 + a1 * a1 : 0.f;
#pragma unroll
            for (int off = 32; off > 0; off >>= 1)
                ss_ += __shfl_xor(ss_, off, 64);
            const float rr = rsqrtf(ss_ / (float)D + eps);
            a0 = a0 * rr * wk0;
            a1 = a1 * rr * wk1;
            const float o0 = a0 * cvec[i] - a1 * svec[i];
            const float o1 = a0 * svec[i] + a1 * cvec[i];
            if (act)
                kp[i][lane] = (unsigned int)__bfloat16_as_ushort(f2bf(o0)) |
                              ((unsigned int)__bfloat16_as_ushort(f2bf(o1)) << 16);
        }
    }
}

int main(int argc, char** argv) {
    const int rounds = argc > 1 ? atoi(argv[1]) : 12;
    // flux single-block shape: [B, S, 3, H, D] qkv, q/k strided views
    const int B = 8, S = 4608, H = 24, D = 128;
    const long rows = (long)B * S * H;
    const long n = (long)B * S * 3 * H * D;
    bf16 *qkv, *w, *o_ref, *o_got;
    float* cs;
    HIP_CHECK(hipMalloc(&qkv, n * 2));
    HIP_CHECK(hipMalloc(&w, 2 * D * 2));
    HIP_CHECK(hipMalloc(&cs, (long)S * (D / 2) * 2 * 4));
    {
        std::vector<unsigned short> h(1 << 20);
        unsigned x = 7;
        for (auto& e : h) {
            x = x * 1664525u + 1013904223u;
            float f = ((x >> 8) / 8388608.0f) * 2.f - 1.f;
            unsigned int bits; __builtin_memcpy(&bits, &f, 4);
            e = (unsigned short)(bits >> 16);
        }
        for (long off = 0; off < n; off += (1 << 20))
            HIP_CHECK(hipMemcpy(qkv + off, h.data(),
                                std::min<long>(1 << 20, n - off) * 2,
                                hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(w, h.data(), 2 * D * 2, hipMemcpyHostToDevice));
        std::vector<float> hc((long)S * (D / 2) * 2);
        for (size_t i = 0; i < hc.size(); i += 2) {
            hc[i] = cosf(0.001f * i);
            hc[i + 1] = sinf(0.001f * i);
        }
        HIP_CHECK(hipMemcpy(cs, hc.data(), hc.size() * 4,
                            hipMemcpyHostToDevice));
    }
    bf16* q = qkv;                 // qkv slot 0
    bf16* k = qkv + (long)H * D;   // qkv slot 1
    const long q_bs = (long)S * 3 * H * D;
    const long q_ss = (long)3 * H * D;
    const long q_hs = D;
    auto run = [&](int rpw) {
        const long waves = (rows + rpw - 1) / rpw;
        const long blocks = (waves * 64 + 255) / 256;
        switch (rpw) {
        case 4:
            hipLaunchKernelGGL(qknr_kernel<4>, dim3((unsigned)blocks),
                               dim3(256), 0, 0, q, k, w, w + D, cs, S, H, D,
                               q_bs, q_hs, q_ss, rows, 1e-6f);
            break;
        case 8:
            hipLaunchKernelGGL(qknr_kernel<8>, dim3((unsigned)blocks),
                               dim3(256), 0, 0, q, k, w, w + D, cs, S, H, D,
                               q_bs, q_hs, q_ss, rows, 1e-6f);
            break;
        default:
            hipLaunchKernelGGL(qknr_kernel<16>, dim3((unsigned)blocks),
                               dim3(256), 0, 0, q, k, w, w + D, cs, S, H, D,
                               q_bs, q_hs, q_ss, rows, 1e-6f);
        }
    };
    // in-place kernel: timing-only comparison (correctness of the shipped
    // RPW=4 form is covered by tests/test_gpu_kernels.py)
    const int RPWS[3] = {4, 8, 16};
    double best[3] = {1e30, 1e30, 1e30};
    run(4);
    HIP_CHECK(hipGetLastError());
    HIP_CHECK(hipDeviceSynchronize());
    const double gb = rows * (double)D * 2 * 2 * 2 / 1e9;  // q+k, r+w
    for (int r = 0; r < rounds; ++r)
        for (int i = 0; i < 3; ++i) {
            hipEvent_t e0, e1;
            HIP_CHECK(hipEventCreate(&e0));
            HIP_CHECK(hipEventCreate(&e1));
            HIP_CHECK(hipEventRecord(e0));
            for (int it = 0; it < 3; ++it) run(RPWS[i]);
            HIP_CHECK(hipEventRecord(e1));
            HIP_CHECK(hipEventSynchronize(e1));
            float ms;
            HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
            if (ms / 3 < best[i]) best[i] = ms / 3;
            (void)hipEventDestroy(e0);
            (void)hipEventDestroy(e1);
        }
    for (int i = 0; i < 3; ++i)
        printf("RPW=%-2d  %8.1f us  %6.2f TB/s\n", RPWS[i], best[i] * 1e3,
               gb / best[i]);
    return 0;
}
