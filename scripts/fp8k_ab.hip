// fp8-emitting elementwise kernels A/B (gfx950): why do gelu_fp8 /
// quant_fp8 / layer_norm_mod_fp8 run at ~2.3-3.0 TB/s while the bf16
// analogs hit 6+?
//
// Variants on the gelu_fp8 shape ([8,4608,12288] bf16 in, e4m3 out):
//   0: shipped structure (grid<=4096, per-block atomicMax+counter tail)
//   1: grid<=16384
//   2: no epilogue at all (amax tracked in registers, never published)
//      -> isolates the same-address atomic chain cost
//   3: epilogue, but only blocks with (blockIdx % 16 == 0) participate
//      in the atomics; others publish into a scratch line that the last
//      atomic block folds in -- 256 atomics instead of 4096
//   4: variant 0 + 16-byte stores (two vec-iters packed per store)
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 fp8k_ab.hip -o fp8k
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <algorithm>

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) short short8;
#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP error %d at %d\n", e, __LINE__); exit(1); } } while (0)

__device__ __forceinline__ float bf2f(bf16 v) { return __bfloat162float(v); }

// v8: 4-deep ILP + CONDITIONAL atomicMax (skip when not a new max) +
// the highest-blockIdx block writes the next scale from the running amax
// (no counter chain; delayed scaling tolerates a fraction-of-a-call lag)
__global__ void gelu_fp8_v8(const bf16* __restrict__ x,
                            unsigned char* __restrict__ out, long total8,
                            float* __restrict__ scale,
                            float* __restrict__ amax_buf,
                            float* __restrict__ scale_used) {
    const long stride = (long)gridDim.x * blockDim.x;
    const short8* xv = reinterpret_cast<const short8*>(x);
    const float s_entry = scale[0];
    const float inv_s = 1.0f / s_entry;
    float local_amax = 0.f;
    const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    for (long i = i0; i < total8; i += stride * 4) {
        short8 v[4];
        long idx[4];
#pragma unroll
        for (int h = 0; h < 4; ++h) {
            idx[h] = i + h * stride;
            if (idx[h] < total8) v[h] = xv[idx[h]];
        }
#pragma unroll
        for (int h = 0; h < 4; ++h) {
            if (idx[h] >= total8) break;
            unsigned char pack[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float f =
                    bf2f(__ushort_as_bfloat16((unsigned short)v[h][j]));
                const float c =
                    0.7978845608028654f * (f + 0.044715f * f * f * f);
                const float En = __builtin_amdgcn_exp2f(
                    -2.8853900817779268f * fabsf(c));
                const float r = 1.f / (1.f + En);
                const float g = f * (c >= 0.f ? r : 1.f - r);
                local_amax = fmaxf(local_amax, fabsf(g));
                const float qv = fminf(fmaxf(g * inv_s, -448.f), 448.f);
                pack[j] = (unsigned char)__hip_cvt_float_to_fp8(
                    qv, __HIP_SATFINITE, __HIP_E4M3);
            }
            *reinterpret_cast<unsigned long long*>(&out[idx[h] * 8]) =
                *reinterpret_cast<unsigned long long*>(pack);
        }
    }
    __shared__ float red[8];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        local_amax = fmaxf(local_amax, __shfl_xor(local_amax, off, 64));
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    if (lane == 0) red[wid] = local_amax;
    __syncthreads();
    if (threadIdx.x == 0) {
        float m = 0.f;
        for (int i = 0; i < (int)(blockDim.x / 64); ++i)
            m = fmaxf(m, red[i]);
        if (m > amax_buf[0])  // plain read; atomic only on a new max
            atomicMax(reinterpret_cast<unsigned int*>(amax_buf),
                      __float_as_uint(m));
        if (blockIdx.x == gridDim.x - 1) {
            __threadfence();
            const float next = fmaxf(amax_buf[0], m) * 0.999f;
            amax_buf[0] = next;
            scale_used[0] = s_entry;
            scale[0] = fmaxf(next / 448.f, 1e-12f);
        }
    }
}

template <int VAR>
__global__ void gelu_fp8_ab(const bf16* __restrict__ x,
                            unsigned char* __restrict__ out, long total8,
                            float* __restrict__ scale,
                            float* __restrict__ amax_buf,
                            float* __restrict__ scratch_amax) {
    const long stride = (long)gridDim.x * blockDim.x;
    const short8* xv = reinterpret_cast<const short8*>(x);
    const float inv_s = 1.0f / scale[0];
    float local_amax = 0.f;
    const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (VAR == 5) {
        // pure 16B-load -> 8B-store passthrough (no gelu, no cvt):
        // the structural ceiling of this access pattern
        for (long i = i0; i < total8; i += stride) {
            short8 v = xv[i];
            unsigned int lo = ((unsigned int)(unsigned short)v[0]) |
                              ((unsigned int)(unsigned short)v[2] << 16);
            unsigned int hi2 = ((unsigned int)(unsigned short)v[4]) |
                               ((unsigned int)(unsigned short)v[6] << 16);
            *reinterpret_cast<unsigned long long*>(&out[i * 8]) =
                (unsigned long long)lo | ((unsigned long long)hi2 << 32);
        }
        return;
    } else if (VAR == 6) {
        // 4-deep manual ILP: 4 independent loads in flight, then compute
        for (long i = i0; i < total8; i += stride * 4) {
            short8 v[4];
            long idx[4];
#pragma unroll
            for (int h = 0; h < 4; ++h) {
                idx[h] = i + h * stride;
                if (idx[h] < total8) v[h] = xv[idx[h]];
            }
#pragma unroll
            for (int h = 0; h < 4; ++h) {
                if (idx[h] >= total8) break;
                unsigned char pack[8];
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    const float f =
                        bf2f(__ushort_as_bfloat16((unsigned short)v[h][j]));
                    const float c =
                        0.7978845608028654f * (f + 0.044715f * f * f * f);
                    const float En = __builtin_amdgcn_exp2f(
                        -2.8853900817779268f * fabsf(c));
                    const float r = 1.f / (1.f + En);
                    const float g = f * (c >= 0.f ? r : 1.f - r);
                    local_amax = fmaxf(local_amax, fabsf(g));
                    const float qv = fminf(fmaxf(g * inv_s, -448.f), 448.f);
                    pack[j] = (unsigned char)__hip_cvt_float_to_fp8(
                        qv, __HIP_SATFINITE, __HIP_E4M3);
                }
                *reinterpret_cast<unsigned long long*>(&out[idx[h] * 8]) =
                    *reinterpret_cast<unsigned long long*>(pack);
            }
        }
        scratch_amax[(blockIdx.x * blockDim.x + threadIdx.x) & 65535] =
            local_amax;
        return;
    } else if (VAR == 7) {
        // gelu compute but STORE BF16 (16B) — isolates the 8B-store cost
        for (long i = i0; i < total8; i += stride) {
            short8 v = xv[i], ovv;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float f =
                    bf2f(__ushort_as_bfloat16((unsigned short)v[j]));
                const float c =
                    0.7978845608028654f * (f + 0.044715f * f * f * f);
                const float En = __builtin_amdgcn_exp2f(
                    -2.8853900817779268f * fabsf(c));
                const float r = 1.f / (1.f + En);
                const float g = f * (c >= 0.f ? r : 1.f - r);
                local_amax = fmaxf(local_amax, fabsf(g));
                ovv[j] = (short)__bfloat16_as_ushort(__float2bfloat16(g));
            }
            *reinterpret_cast<short8*>(&out[i * 16]) = ovv;
        }
        scratch_amax[(blockIdx.x * blockDim.x + threadIdx.x) & 65535] =
            local_amax;
        return;
    }
    if (VAR == 4) {
        // two vec-iters per loop -> one 16-byte store
        for (long i = i0 * 2; i + 1 < total8 * 1; i += stride * 2) {
            unsigned long long pk[2];
#pragma unroll
            for (int h = 0; h < 2; ++h) {
                short8 v = xv[i + h * 1];
                unsigned char pack[8];
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    const float f =
                        bf2f(__ushort_as_bfloat16((unsigned short)v[j]));
                    const float c =
                        0.7978845608028654f * (f + 0.044715f * f * f * f);
                    const float En = __builtin_amdgcn_exp2f(
                        -2.8853900817779268f * fabsf(c));
                    const float r = 1.f / (1.f + En);
                    const float g = f * (c >= 0.f ? r : 1.f - r);
                    local_amax = fmaxf(local_amax, fabsf(g));
                    const float qv = fminf(fmaxf(g * inv_s, -448.f), 448.f);
                    pack[j] = (unsigned char)__hip_cvt_float_to_fp8(
                        qv, __HIP_SATFINITE, __HIP_E4M3);
                }
                pk[h] = *reinterpret_cast<unsigned long long*>(pack);
            }
            *reinterpret_cast<unsigned long long*>(&out[i * 8]) = pk[0];
            *reinterpret_cast<unsigned long long*>(&out[i * 8 + 8]) = pk[1];
        }
    } else {
        for (long i = i0; i < total8; i += stride) {
            short8 v = xv[i];
            unsigned char pack[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float f =
                    bf2f(__ushort_as_bfloat16((unsigned short)v[j]));
                const float c =
                    0.7978845608028654f * (f + 0.044715f * f * f * f);
                const float En = __builtin_amdgcn_exp2f(
                    -2.8853900817779268f * fabsf(c));
                const float r = 1.f / (1.f + En);
                const float g = f * (c >= 0.f ? r : 1.f - r);
                local_amax = fmaxf(local_amax, fabsf(g));
                const float qv = fminf(fmaxf(g * inv_s, -448.f), 448.f);
                pack[j] = (unsigned char)__hip_cvt_float_to_fp8(
                    qv, __HIP_SATFINITE, __HIP_E4M3);
            }
            *reinterpret_cast<unsigned long long*>(&out[i * 8]) =
                *reinterpret_cast<unsigned long long*>(pack);
        }
    }
    if (VAR == 2 || VAR >= 5) {
        // no ATOMIC epilogue — but the amax chain must stay live (a sign
        // test lets the compiler DCE the whole fmax chain): plain store
        scratch_amax[(blockIdx.x * blockDim.x + threadIdx.x) & 65535] =
            local_amax;
        return;
    }
    __shared__ float red[8];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        local_amax = fmaxf(local_amax, __shfl_xor(local_amax, off, 64));
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    if (lane == 0) red[wid] = local_amax;
    __syncthreads();
    if (threadIdx.x == 0) {
        float m = 0.f;
        for (int i = 0; i < (int)(blockDim.x / 64); ++i)
            m = fmaxf(m, red[i]);
        if (VAR == 3) {
            // sparse atomics: publish to a private scratch line; every
            // 16th block folds its group's scratch via atomics
            scratch_amax[blockIdx.x] = m;
            __threadfence();
            const unsigned int done = atomicAdd(
                reinterpret_cast<unsigned int*>(&amax_buf[1]), 1u);
            if (done == gridDim.x - 1) {
                float g = 0.f;
                for (unsigned int i = 0; i < gridDim.x; ++i)
                    g = fmaxf(g, scratch_amax[i]);
                reinterpret_cast<unsigned int*>(amax_buf)[1] = 0u;
                const float next = fmaxf(amax_buf[0] * 0.999f, g * 0.999f);
                amax_buf[0] = next;
                scale[0] = fmaxf(next / 448.f, 1e-12f);
            }
        } else {
            atomicMax(reinterpret_cast<unsigned int*>(amax_buf),
                      __float_as_uint(m));
            __threadfence();
            const unsigned int done = atomicAdd(
                reinterpret_cast<unsigned int*>(&amax_buf[1]), 1u);
            if (done == gridDim.x - 1) {
                reinterpret_cast<unsigned int*>(amax_buf)[1] = 0u;
                const float next = amax_buf[0] * 0.999f;
                amax_buf[0] = next;
                scale[0] = fmaxf(next / 448.f, 1e-12f);
            }
        }
    }
}

int main(int argc, char** argv) {
    const int rounds = argc > 1 ? atoi(argv[1]) : 10;
    const long n = 8L * 4608 * 12288;  // the gelu_fp8 shape
    const long total8 = n / 8;
    bf16* x;
    unsigned char* o;
    float *scale, *amax, *scratch;
    HIP_CHECK(hipMalloc(&x, n * 2));
    HIP_CHECK(hipMalloc(&o, n * 2));  // v7 stores bf16 (2 bytes/elem)
    HIP_CHECK(hipMalloc(&scale, 4));
    HIP_CHECK(hipMalloc(&amax, 8));
    HIP_CHECK(hipMalloc(&scratch, 4 * 65536));
    {
        std::vector<unsigned short> h(1 << 20);
        unsigned s = 99;
        for (auto& e : h) {
            s = s * 1664525u + 1013904223u;
            float f = ((s >> 8) / 8388608.0f) * 2.f - 1.f;
            unsigned int bits; __builtin_memcpy(&bits, &f, 4);
            e = (unsigned short)(bits >> 16);
        }
        for (long off = 0; off < n; off += (1 << 20))
            HIP_CHECK(hipMemcpy(x + off, h.data(),
                                std::min<long>(1 << 20, n - off) * 2,
                                hipMemcpyHostToDevice));
        float one = 0.01f;
        HIP_CHECK(hipMemcpy(scale, &one, 4, hipMemcpyHostToDevice));
        HIP_CHECK(hipMemset(amax, 0, 8));
    }
    const double gb = (n * 2 + n) / 1e9;      // 16B read + 8B write
    const double gb16 = (n * 2 + n * 2) / 1e9;  // v7: 16B read + 16B write
    auto run = [&](int var, int grid) {
        dim3 g(grid), b(256);
        switch (var) {
        case 0: hipLaunchKernelGGL(gelu_fp8_ab<0>, g, b, 0, 0, x, o, total8,
                                   scale, amax, scratch); break;
        case 1: hipLaunchKernelGGL(gelu_fp8_ab<1>, g, b, 0, 0, x, o, total8,
                                   scale, amax, scratch); break;
        case 2: hipLaunchKernelGGL(gelu_fp8_ab<2>, g, b, 0, 0, x, o, total8,
                                   scale, amax, scratch); break;
        case 3: hipLaunchKernelGGL(gelu_fp8_ab<3>, g, b, 0, 0, x, o, total8,
                                   scale, amax, scratch); break;
        case 5: hipLaunchKernelGGL(gelu_fp8_ab<5>, g, b, 0, 0, x, o, total8,
                                   scale, amax, scratch); break;
        case 6: hipLaunchKernelGGL(gelu_fp8_ab<6>, g, b, 0, 0, x, o, total8,
                                   scale, amax, scratch); break;
        case 7: hipLaunchKernelGGL(gelu_fp8_ab<7>, g, b, 0, 0, x, o, total8,
                                   scale, amax, scratch); break;
        case 8: hipLaunchKernelGGL(gelu_fp8_v8, g, b, 0, 0, x, o, total8,
                                   scale, amax, scratch); break;
        default: hipLaunchKernelGGL(gelu_fp8_ab<4>, g, b, 0, 0, x, o, total8,
                                    scale, amax, scratch); break;
        }
    };
    struct V { int var; int grid; const char* name; };
    V vs[] = {{0, 4096, "v0 shipped structure"},
              {2, 4096, "v2 no-atomics (amax live)"},
              {6, 4096, "v6 ILP4 no-atomics"},
              {8, 4096, "v8 ILP4+cond-atomic+heur"},
              {5, 4096, "v5 passthrough"}};
    const int NV = sizeof(vs) / sizeof(vs[0]);
    double best[8];
    for (auto& t : best) t = 1e30;
    run(0, 4096);
    HIP_CHECK(hipGetLastError());
    HIP_CHECK(hipDeviceSynchronize());
    for (int r = 0; r < rounds; ++r)
        for (int i = 0; i < NV; ++i) {
            hipEvent_t e0, e1;
            HIP_CHECK(hipEventCreate(&e0));
            HIP_CHECK(hipEventCreate(&e1));
            HIP_CHECK(hipEventRecord(e0));
            for (int it = 0; it < 3; ++it) run(vs[i].var, vs[i].grid);
            HIP_CHECK(hipEventRecord(e1));
            HIP_CHECK(hipEventSynchronize(e1));
            float ms;
            HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
            if (ms / 3 < best[i]) best[i] = ms / 3;
            (void)hipEventDestroy(e0);
            (void)hipEventDestroy(e1);
        }
    for (int i = 0; i < NV; ++i)
        printf("%-26s %8.1f us  %6.2f TB/s\n", vs[i].name, best[i] * 1e3,
               (vs[i].var == 7 ? gb16 : gb) / best[i]);
    return 0;
}
