// Standalone within-probe A/B harness for the v4 attention kernel (gfx950).
// Variants share one process and interleave rounds (guide §5.4 rule 24) so
// deltas are reliable despite cross-run/DVFS noise.
//
// Variants:
//   0: shipped v4
//   1: tr_b16 V path (guide T10): V stored ROW-major in LDS with a
//      40-granule row stride (≡8 mod 32 -> conflict-free tr gather) and
//      a (row&8)<<1 column XOR; written with plain b128 stores (the 16
//      scalar V^T stores per thread disappear); PV A-fragments read with
//      ds_read_b64_tr_b16 (2 per (chunk, tile), base VGPR + immediates).
//      Gather semantics verified empirically (scripts/tr_probe.hip):
//      within a 16-lane group, lane L reg j = element (L&3) of the
//      granule addressed by lane (L>>2)+4j.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 attn_ab.hip -o attn_ab
// Run:   ./attn_ab [rounds]

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <algorithm>
#include <cmath>

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
#define LDS_P __attribute__((address_space(3)))
#define PA_DEV __device__ __forceinline__
#define PA_LOG2E 1.4426950408889634f

PA_DEV float bf2f(bf16 v) { return __bfloat162float(v); }
PA_DEV bf16 f2bf(float v) { return __float2bfloat16(v); }
PA_DEV f32x16 mfma32x32x16(bf16x8 a, bf16x8 b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}
PA_DEV unsigned int cvt_pk_bf16(float lo, float hi) {
    unsigned int r;
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
    return r;
}

template <int D, int VAR>
__global__ __launch_bounds__(512, 2) void attn_ab_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out,
    int S, int Sk, float scale, int H) {
    constexpr int KVBLK = 64;
    constexpr int WAVES = 8;
    constexpr int SUPER = 1;
    constexpr int DEPTH = 1;
    constexpr bool NT_IO = false;
    constexpr bool STATIC_PRIO = false;
    constexpr bool TRV = true;         // tr_b16 V path (shipped baseline)
    constexpr bool DEFER = (VAR == 1); // T13 defer-max RESCALE_THRESHOLD
    constexpr float DEFER_THR = 8.0f;  // exp2 domain: P bounded by 2^8
    constexpr int THREADS = WAVES * 64;
    constexpr int KPAD = D + 8;
    constexpr int VPAD = KVBLK + 8;
    constexpr int VROW = 160;          // TRV row stride (elements): 40 granules
    constexpr int KK = D / 16;
    constexpr int NV = D / 32;
    constexpr int KVECS = (SUPER * KVBLK * D) / (8 * THREADS);
    constexpr bool RAW_EXP = true;
    constexpr bool PTR_INC = false;
    constexpr bool TAIL_SPEC = false;
    constexpr bool KT_ILV = false;
    constexpr bool UNCOND_ALPHA = false;
    constexpr bool NO_PRIO = false;
    constexpr bool VPAIR = false;

    __shared__ bf16 k_lds[SUPER * KVBLK * KPAD];
    __shared__ bf16 v_lds[TRV ? (KVBLK * VROW) : (SUPER * D * VPAD)];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l32 = lane & 31;
    const int hi = lane >> 5;
    // tr_b16 per-lane base: this lane supplies the granule
    // (key = hi*8 + (p>>2) [+16c+4rd], col = 32n + 16*((G&1)^hi) + 4*(p&3))
    const int trb = (hi * 8 + ((lane & 15) >> 2)) * VROW +
                    16 * (((lane >> 4) & 1) ^ hi) + 4 * (lane & 3);
    if (STATIC_PRIO &&
        __builtin_amdgcn_readfirstlane(threadIdx.x) >= (WAVES / 2) * 64)
        __builtin_amdgcn_s_setprio(1);

    // XCD-affine decode (1-D grid)
    const int nq = (S + WAVES * 32 - 1) / (WAVES * 32);
    const long id = blockIdx.x;
    const long bh = (id & 7) + 8 * ((id >> 3) / nq);
    const int qtile = (int)((id >> 3) % nq);
    const long b = bh / H;
    const int h = (int)(bh % H);
    const int q0 = qtile * (WAVES * 32) + wid * 32;
    const int ss = H * D;  // packed BSHD row stride

    const bf16* qp = q + (b * (long)S + 0) * ss + (long)h * D;
    const bf16* kp = k + (b * (long)Sk + 0) * ss + (long)h * D;
    const bf16* vp = v + (b * (long)Sk + 0) * ss + (long)h * D;
    bf16* op = out + (b * (long)S + 0) * ss + (long)h * D;

    bf16x8 qfrag[KK];
    {
        const int row = q0 + l32;
        const int rr = row < S ? row : S - 1;
#pragma unroll
        for (int kk = 0; kk < KK; ++kk) {
            const bf16x8* src = reinterpret_cast<const bf16x8*>(
                qp + (long)rr * ss + kk * 16 + hi * 8);
            qfrag[kk] = NT_IO ? __builtin_nontemporal_load(src) : *src;
        }
    }

    f32x16 o_acc[NV];
#pragma unroll
    for (int n = 0; n < NV; ++n)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[n][r] = 0.f;
    float m_run = -1e30f, l_run = 0.f;
    const float scale2 = scale * PA_LOG2E;

    bf16x8 kreg[DEPTH][KVECS], vreg[DEPTH][KVECS];
    const bf16* kptr[KVECS];
    const bf16* vptr[KVECS];
    int lrow[KVECS];
    if (PTR_INC) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = tid + i * THREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            lrow[i] = row;
            kptr[i] = kp + (long)row * ss + col;
            vptr[i] = vp + (long)row * ss + col;
        }
    }

    auto issue_tile_loads = [&](int kv0, int slot) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            {
                const int idx = tid + i * THREADS;
                const int row = idx / (D / 8);
                const int col = (idx % (D / 8)) * 8;
                const int src = kv0 + row;
                if (src < Sk) {
                    kreg[slot][i] = *reinterpret_cast<const bf16x8*>(kp + (long)src * ss + col);
                    vreg[slot][i] = *reinterpret_cast<const bf16x8*>(vp + (long)src * ss + col);
                } else {
                    kreg[slot][i] = bf16x8{0,0,0,0,0,0,0,0};
                    vreg[slot][i] = bf16x8{0,0,0,0,0,0,0,0};
                }
            }
        }
    };
    auto write_tile_lds = [&](int slot) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = tid + i * THREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            const int sub = row / KVBLK;          // 64-key sub-tile id
            const int srow = row % KVBLK;
            *reinterpret_cast<bf16x8*>(&k_lds[row * KPAD + col]) = kreg[slot][i];
            if (TRV) {
                // row-major V, b128 store; column XOR by (row&8)<<1 keeps the
                // tr gather conflict-free under either half-wave pairing
                *reinterpret_cast<bf16x8*>(
                    &v_lds[row * VROW + (col ^ ((row & 8) << 1))]) =
                    vreg[slot][i];
            } else if (VPAIR && D == 128) {
                // pair keys (row, row^1) via shfl_xor(16): even-key threads
                // write dims col..col+3 as b32, odd-key threads col+4..col+7
                const bool even = ((lane >> 4) & 1) == 0;
#pragma unroll
                for (int jj = 0; jj < 4; ++jj) {
                    const int j = even ? jj : jj + 4;
                    const unsigned short mine =
                        (unsigned short)vreg[slot][i][j];
                    const unsigned short partner = (unsigned short)__shfl_xor(
                        (int)(unsigned short)vreg[slot][i][j ^ 4], 16, 64);
                    const unsigned short lo_key = even ? mine : partner;
                    const unsigned short hi_key = even ? partner : mine;
                    const int dim = col + j;
                    const int key0 = row & ~1;
                    const int key_swz =
                        (((key0 >> 3) ^ ((dim >> 3) & 7)) << 3) | (key0 & 7);
                    *reinterpret_cast<unsigned int*>(
                        &v_lds[dim * VPAD + key_swz]) =
                        (unsigned int)lo_key | ((unsigned int)hi_key << 16);
                }
            } else {
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    const int dim = col + j;
                    const int key_swz =
                        (((srow >> 3) ^ ((dim >> 3) & 7)) << 3) | (srow & 7);
                    v_lds[(sub * D + dim) * VPAD + key_swz] =
                        __ushort_as_bfloat16((unsigned short)vreg[slot][i][j]);
                }
            }
        }
    };

    auto exp2x = [](float x) { return __builtin_amdgcn_exp2f(x); };
    (void)RAW_EXP; (void)PTR_INC; (void)TAIL_SPEC;

    auto tile = [&](int kv0, bool mask, int sub) {
        f32x16 st[2];
#pragma unroll
        for (int kt = 0; kt < 2; ++kt)
#pragma unroll
            for (int r = 0; r < 16; ++r) st[kt][r] = 0.f;
        if (!NO_PRIO) __builtin_amdgcn_s_setprio(1);
        if (KT_ILV) {
#pragma unroll
            for (int kk = 0; kk < KK; ++kk)
#pragma unroll
                for (int kt = 0; kt < 2; ++kt) {
                    bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
                        &k_lds[(kt * 32 + l32) * KPAD + kk * 16 + hi * 8]);
                    st[kt] = mfma32x32x16(afrag, qfrag[kk], st[kt]);
                }
        } else {
#pragma unroll
            for (int kt = 0; kt < 2; ++kt)
#pragma unroll
                for (int kk = 0; kk < KK; ++kk) {
                    bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
                        &k_lds[(kt * 32 + l32) * KPAD + kk * 16 + hi * 8]);
                    st[kt] = mfma32x32x16(afrag, qfrag[kk], st[kt]);
                }
        }
        if (!NO_PRIO) __builtin_amdgcn_s_setprio(0);
        float mx = -1e30f;
#pragma unroll
        for (int kt = 0; kt < 2; ++kt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                float sv = st[kt][r] * scale2;
                if (mask) {
                    const int key =
                        kv0 + kt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
                    sv = (key < Sk) ? sv : -1e30f;
                }
                st[kt][r] = sv;
                mx = fmaxf(mx, sv);
            }
        mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
        // T13: if the max grew by < THR, keep the old max — alpha == 1
        // exactly, so the O rescale below is skipped; P is then bounded by
        // 2^THR instead of 1 (bf16-accum headroom covers it).
        const float mnew = DEFER ? (mx <= m_run + DEFER_THR ? m_run : mx)
                                 : fmaxf(m_run, mx);
        const float alpha = exp2x(m_run - mnew);
        m_run = mnew;
        float ps = 0.f;
#pragma unroll
        for (int kt = 0; kt < 2; ++kt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const float pv_ = exp2x(st[kt][r] - mnew);
                st[kt][r] = pv_;
                ps += pv_;
            }
        ps += __shfl_xor(ps, 32, 64);
        l_run = l_run * alpha + ps;
        if (UNCOND_ALPHA) {
#pragma unroll
            for (int n = 0; n < NV; ++n)
#pragma unroll
                for (int r = 0; r < 16; ++r) o_acc[n][r] *= alpha;
        } else if (alpha != 1.f) {
#pragma unroll
            for (int n = 0; n < NV; ++n)
#pragma unroll
                for (int r = 0; r < 16; ++r) o_acc[n][r] *= alpha;
        }
        bf16x8 pfrag[4];
#pragma unroll
        for (int c = 0; c < 4; ++c) {
            const f32x16& sv = st[c >> 1];
            const int rb = 8 * (c & 1);
            unsigned int w0 = cvt_pk_bf16(sv[rb + 0], sv[rb + 1]);
            unsigned int w1 = cvt_pk_bf16(sv[rb + 2], sv[rb + 3]);
            unsigned int w2 = cvt_pk_bf16(sv[rb + 4], sv[rb + 5]);
            unsigned int w3 = cvt_pk_bf16(sv[rb + 6], sv[rb + 7]);
            auto r02 = __builtin_amdgcn_permlane32_swap(w0, w2, false, false);
            auto r13 = __builtin_amdgcn_permlane32_swap(w1, w3, false, false);
            unsigned int d[4] = {(unsigned int)r02[0], (unsigned int)r13[0],
                                 (unsigned int)r02[1], (unsigned int)r13[1]};
            pfrag[c] = *reinterpret_cast<bf16x8*>(d);
        }
        if (!NO_PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int c = 0; c < 4; ++c) {
#pragma unroll
            for (int n = 0; n < NV; ++n) {
                bf16x8 va;
                if (TRV) {
                    s16x4 alo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                        (LDS_P s16x4*)&v_lds[trb + c * (16 * VROW) + n * 32]);
                    s16x4 ahi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                        (LDS_P s16x4*)&v_lds[trb + c * (16 * VROW) +
                                             4 * VROW + n * 32]);
                    va = __builtin_shufflevector(alo, ahi,
                                                 0, 1, 2, 3, 4, 5, 6, 7);
                } else {
                    const int dim = n * 32 + l32;
                    const int gsw = ((2 * c + hi) ^ ((dim >> 3) & 7)) << 3;
                    va = *reinterpret_cast<const bf16x8*>(
                        &v_lds[(sub * D + dim) * VPAD + gsw]);
                }
                o_acc[n] = mfma32x32x16(va, pfrag[c], o_acc[n]);
            }
        }
        if (!NO_PRIO) __builtin_amdgcn_s_setprio(0);
    };

    const int n_tiles = (Sk + KVBLK - 1) / KVBLK;
    issue_tile_loads(0, 0);
    if (DEPTH == 2 && n_tiles > 1) issue_tile_loads(KVBLK, 1);
    for (int t = 0; t < n_tiles; ++t) {
        __syncthreads();
        write_tile_lds(t % DEPTH);
        __syncthreads();
        if (t + DEPTH < n_tiles || (DEPTH == 1 && t + 1 < n_tiles))
            issue_tile_loads((t + DEPTH) * KVBLK, (t + DEPTH) % DEPTH);
        tile(t * KVBLK, true, 0);
    }

    const int row = q0 + l32;
    if (row < S) {
        const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
        for (int n = 0; n < NV; ++n)
#pragma unroll
            for (int r2 = 0; r2 < 4; ++r2) {
                const int dim0 = n * 32 + 8 * r2 + 4 * hi;
                unsigned short pack[4];
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    pack[j] = __bfloat16_as_ushort(
                        f2bf(o_acc[n][r2 * 4 + j] * inv_l));
                unsigned long long* dst = reinterpret_cast<unsigned long long*>(
                    op + (long)row * ss + dim0);
                const unsigned long long pv =
                    *reinterpret_cast<unsigned long long*>(pack);
                if (NT_IO)
                    __builtin_nontemporal_store(pv, dst);
                else
                    *dst = pv;
            }
    }
}

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP error %d at %d\n", e, __LINE__); exit(1); } } while (0)

int main(int argc, char** argv) {
    const int rounds = argc > 1 ? atoi(argv[1]) : 8;
    struct Shape { int B, H, S, D; const char* name; };
    Shape shapes[3] = {{8, 24, 4608, 128, "flux"}, {1, 8, 30720, 128, "long"},
                       {8, 38, 4250, 64, "sd3-d64"}};
    for (auto& sh : shapes) {
        const int B = sh.B, H = sh.H, S = sh.S, D = sh.D;
        const long n = (long)B * S * H * D;
        bf16 *q, *k, *v, *o;
        HIP_CHECK(hipMalloc(&q, n * 2));
        HIP_CHECK(hipMalloc(&k, n * 2));
        HIP_CHECK(hipMalloc(&v, n * 2));
        HIP_CHECK(hipMalloc(&o, n * 2));
        {
            std::vector<unsigned short> h(1 << 20);
            unsigned x = 12345;
            for (auto& e : h) {
                x = x * 1664525u + 1013904223u;
                float f = ((x >> 8) / 8388608.0f) * 2.f - 1.f;
                unsigned int bits; __builtin_memcpy(&bits, &f, 4);
                e = (unsigned short)(bits >> 16);
            }
            for (long off = 0; off < n; off += (1 << 20)) {
                long len = std::min<long>(1 << 20, n - off);
                HIP_CHECK(hipMemcpy(q + off, h.data(), len * 2, hipMemcpyHostToDevice));
                HIP_CHECK(hipMemcpy(k + off, h.data(), len * 2, hipMemcpyHostToDevice));
                HIP_CHECK(hipMemcpy(v + off, h.data(), len * 2, hipMemcpyHostToDevice));
            }
        }
        const float scale = 1.0f / sqrtf((float)D);
        const double tf = 4.0 * B * H * (double)S * S * D / 1e12;
        auto run = [&](int var) {
            dim3 grid(((S + 255) / 256) * B * H), blk(512);
            if (D == 64) {
                if (var == 0)
                    hipLaunchKernelGGL((attn_ab_kernel<64, 0>), grid, blk, 0,
                                       0, q, k, v, o, S, S, scale, H);
                else
                    hipLaunchKernelGGL((attn_ab_kernel<64, 1>), grid, blk, 0,
                                       0, q, k, v, o, S, S, scale, H);
            } else if (var == 0)
                hipLaunchKernelGGL((attn_ab_kernel<128, 0>), grid, blk, 0, 0,
                                   q, k, v, o, S, S, scale, H);
            else
                hipLaunchKernelGGL((attn_ab_kernel<128, 1>), grid, blk, 0, 0,
                                   q, k, v, o, S, S, scale, H);
        };
        // correctness gate: variants must agree with variant 0
        std::vector<unsigned short> ref(4096), got(4096);
        run(0); HIP_CHECK(hipGetLastError());
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipMemcpy(ref.data(), o, 4096 * 2, hipMemcpyDeviceToHost));
        for (int var = 1; var < 2; ++var) {
            HIP_CHECK(hipMemset(o, 0, n * 2));
            run(var); HIP_CHECK(hipGetLastError());
            HIP_CHECK(hipDeviceSynchronize());
            HIP_CHECK(hipMemcpy(got.data(), o, 4096 * 2, hipMemcpyDeviceToHost));
            int bad = 0;
            for (int i = 0; i < 4096; ++i) {
                float a, b2;
                unsigned int ua = (unsigned int)ref[i] << 16,
                             ub = (unsigned int)got[i] << 16;
                __builtin_memcpy(&a, &ua, 4);
                __builtin_memcpy(&b2, &ub, 4);
                if (fabsf(a - b2) > 0.05f + 0.05f * fabsf(a)) ++bad;
            }
            if (bad) printf("VARIANT %d WRONG: %d/4096 mismatches\n", var, bad);
        }
        double best[2] = {1e30, 1e30};
        for (int r = 0; r < rounds; ++r)
            for (int var = 0; var < 2; ++var) {
                hipEvent_t e0, e1;
                HIP_CHECK(hipEventCreate(&e0));
                HIP_CHECK(hipEventCreate(&e1));
                HIP_CHECK(hipEventRecord(e0));
                for (int it = 0; it < 3; ++it) run(var);
                HIP_CHECK(hipEventRecord(e1));
                HIP_CHECK(hipEventSynchronize(e1));
                float ms;
                HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
                if (ms / 3.0 < best[var]) best[var] = ms / 3.0;
                (void)hipEventDestroy(e0);
                (void)hipEventDestroy(e1);
            }
        printf("%s  kv64 %7.3f ms (%6.1f TF)   kv128 %7.3f ms (%6.1f TF)\n",
               sh.name, best[0], tf / best[0] * 1e3, best[1], tf / best[1] * 1e3);
        (void)hipFree(q); (void)hipFree(k); (void)hipFree(v); (void)hipFree(o);
    }
    return 0;
}
