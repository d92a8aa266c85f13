// v5 deep-pipelined attention schedule A/B harness (gfx950).
//
// Round-1 verdict: v4 sits AT the 2-barrier structural ceiling (~900 TF
// flux). The way past (guide §5.5 regime gate) is a deeper pipeline:
// double-buffered LDS so the K/V stage for tile t+1 overlaps compute of
// tile t with ONE barrier per tile instead of two, stores interleaved
// into the compute stream where the target pipe is idle.
//
// Variants (within-probe interleaved, guide rule 24):
//   0: shipped v4 (single LDS buffer, 2 barriers/tile)
//   1: v5a — double-buffered LDS, 1 barrier/tile; store order:
//      QK^T -> K-store -> softmax -> V-store -> next-loads -> Pcvt -> PV
//   2: v5b — double-buffered LDS, 1 barrier/tile; stores first:
//      K+V-store -> QK^T -> softmax -> next-loads -> Pcvt -> PV
//
// Correctness gates (guide rules 24/26): full-tensor vs var0 on a tail
// shape at D=128 and D=64, a K-row spike forcing late rescale, and a
// bitwise double-run race screen per variant.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 attn_v5.hip -o attn_v5
// Run:   ./attn_v5 [rounds]

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <algorithm>
#include <cmath>

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(4))) int i32x4;
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

// VAR: 0 = v4 (DBUF 1, 2 barriers); 1 = v5a; 2 = v5b;
//      3 = v5a + MX-fp8 QK^T (mfma_scale_f32_32x32x64, K/Q quantized to
//          e4m3 with per-tensor scales folded into the softmax scale;
//          the A/B byte->k map is freely chosen — it cancels as long as
//          BOTH operands use the same packing, verified by mx_probe)
//      4 = 4-slot LDS ring, ONE barrier every TWO tiles (writes land two
//          tiles ahead; wave skew bounded by the ring depth). LDS 151.6 KB
//          -> 1 block/CU: tests whether intra-block desync beats the
//          co-resident second block.
//      5 = cross-tile stagger: iter t runs QK^T(t) then PV(t-1) back to
//          back (matrix beside matrix), softmax(t)+stores in the second
//          barrier segment — barrier pairs the MFMA batches against the
//          partner wave's memory segment (MI355X_MICROARCH item 5 shape).
template <int D, int VAR>
__global__ __launch_bounds__(512, 2) void attn_v5_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out,
    int S, int Sk, float scale, int H, float qk_inv_s, float qk_s2) {
    constexpr int KVBLK = 64;
    constexpr int WAVES = 8;
    constexpr int THREADS = WAVES * 64;
    constexpr int KPAD = D + 8;
    constexpr int VROW = 160;          // 40 granules ≡ 8 mod 32
    constexpr int KK = D / 16;
    constexpr int NV = D / 32;
    constexpr int KVECS = (KVBLK * D) / (8 * THREADS);
    constexpr int DBUF = (VAR == 0) ? 1 : (VAR == 4 ? 4 : 2);
    constexpr bool FP8QK = (VAR == 3);
    // var7 = v5a + STATIC priority for the younger dispatch half instead
    // of per-segment setprio flips (MI355X_MICROARCH §Two waves per SIMD
    // item 4: the second-dispatched half loses arbitration on every
    // segment; one setprio before the loop, no flips).
    constexpr bool STATIC_PRIO = (VAR == 7);
    constexpr int K8ROW = D + 16;      // fp8 K image row stride (16B-aligned
                                       // b128 reads, conflict-free 36-dw rows)
    constexpr int NMX = D / 64;        // MX MFMAs per 32-key tile

    __shared__ bf16 k_lds[FP8QK ? 1 : (DBUF * KVBLK * KPAD)];
    __shared__ unsigned char k8_lds[FP8QK ? (DBUF * KVBLK * K8ROW) : 1];
    __shared__ bf16 v_lds[DBUF * KVBLK * VROW];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l32 = lane & 31;
    const int hi = lane >> 5;
    const int trb = (hi * 8 + ((lane & 15) >> 2)) * VROW +
                    16 * (((lane >> 4) & 1) ^ hi) + 4 * (lane & 3);
    if (STATIC_PRIO &&
        __builtin_amdgcn_readfirstlane(threadIdx.x) >= (WAVES / 2) * 64)
        __builtin_amdgcn_s_setprio(1);

    const int nq = (S + WAVES * 32 - 1) / (WAVES * 32);
    const long id = blockIdx.x;
    const long bh = (id & 7) + 8 * ((id >> 3) / nq);
    const int qtile = (int)((id >> 3) % nq);
    const long b = bh / H;
    const int h = (int)(bh % H);
    const int q0 = qtile * (WAVES * 32) + wid * 32;
    const int ss = H * D;

    const bf16* qp = q + (b * (long)S + 0) * ss + (long)h * D;
    const bf16* kp = k + (b * (long)Sk + 0) * ss + (long)h * D;
    const bf16* vp = v + (b * (long)Sk + 0) * ss + (long)h * D;
    bf16* op = out + (b * (long)S + 0) * ss + (long)h * D;

    bf16x8 qfrag[KK];
    {
        const int row = q0 + l32;
        const int rr = row < S ? row : S - 1;
#pragma unroll
        for (int kk = 0; kk < KK; ++kk)
            qfrag[kk] = *reinterpret_cast<const bf16x8*>(
                qp + (long)rr * ss + kk * 16 + hi * 8);
    }
    // fp8 Q (B-operand of the MX QK^T): lane holds col=l32 (its q-row),
    // byte j of instruction m = dim m*64 + hi*32 + j — the SAME byte->k
    // map the K image uses (consistency is all the hardware requires).
    i32x8 qf8[FP8QK ? NMX : 1];
    if constexpr (FP8QK) {
        const int row = q0 + l32;
        const int rr = row < S ? row : S - 1;
#pragma unroll
        for (int m = 0; m < NMX; ++m) {
            unsigned char bytes[32];
#pragma unroll
            for (int j2 = 0; j2 < 4; ++j2) {
                bf16x8 v8 = *reinterpret_cast<const bf16x8*>(
                    qp + (long)rr * ss + m * 64 + hi * 32 + j2 * 8);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    float f = bf2f(__ushort_as_bfloat16(
                                  (unsigned short)v8[j])) * qk_inv_s;
                    bytes[j2 * 8 + j] = (unsigned char)__hip_cvt_float_to_fp8(
                        f, __HIP_SATFINITE, __HIP_E4M3);
                }
            }
            qf8[m] = *reinterpret_cast<i32x8*>(bytes);
        }
    }

    f32x16 o_acc[NV];
#pragma unroll
    for (int n = 0; n < NV; ++n)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[n][r] = 0.f;
    float m_run = -1e30f, l_run = 0.f;
    // fp8 mode: scores come back in quantized units; the joint q/k scale
    // (squared) folds into the softmax scale — zero extra VALU per element.
    const float scale2 = (FP8QK ? qk_s2 : 1.f) * scale * PA_LOG2E;

    // var4 keeps TWO tiles in flight in registers (slot = tile % 2)
    constexpr int RSLOTS = (VAR == 4) ? 2 : 1;
    bf16x8 kreg_s[RSLOTS][KVECS], vreg_s[RSLOTS][KVECS];
#define kreg kreg_s[0]
#define vreg vreg_s[0]
    auto issue_tile_loads_slot = [&](int kv0, int slot) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = tid + i * THREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            const int src = kv0 + row;
            if (src < Sk) {
                kreg_s[slot][i] = *reinterpret_cast<const bf16x8*>(
                    kp + (long)src * ss + col);
                vreg_s[slot][i] = *reinterpret_cast<const bf16x8*>(
                    vp + (long)src * ss + col);
            } else {
                kreg_s[slot][i] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
                vreg_s[slot][i] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
            }
        }
    };
    auto issue_tile_loads = [&](int kv0) { issue_tile_loads_slot(kv0, 0); };
    auto write_k_slot = [&](int buf, int slot) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = tid + i * THREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            if constexpr (FP8QK) {
                // quantize K during the stage: e4m3 bytes, row-major image
                unsigned char pack[8];
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    float f = bf2f(__ushort_as_bfloat16(
                                  (unsigned short)kreg_s[slot][i][j])) *
                              qk_inv_s;
                    pack[j] = (unsigned char)__hip_cvt_float_to_fp8(
                        f, __HIP_SATFINITE, __HIP_E4M3);
                }
                *reinterpret_cast<unsigned long long*>(
                    &k8_lds[buf * (KVBLK * K8ROW) + row * K8ROW + col]) =
                    *reinterpret_cast<unsigned long long*>(pack);
            } else {
                *reinterpret_cast<bf16x8*>(
                    &k_lds[buf * (KVBLK * KPAD) + row * KPAD + col]) =
                    kreg_s[slot][i];
            }
        }
    };
    auto write_v_slot = [&](int buf, int slot) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = tid + i * THREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            *reinterpret_cast<bf16x8*>(
                &v_lds[buf * (KVBLK * VROW) + row * VROW +
                       (col ^ ((row & 8) << 1))]) = vreg_s[slot][i];
        }
    };
    auto write_k_lds = [&](int buf) { write_k_slot(buf, 0); };
    auto write_v_lds = [&](int buf) { write_v_slot(buf, 0); };

    auto qk_half = [&](int buf, f32x16* st) {
        if constexpr (!STATIC_PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
            for (int r = 0; r < 16; ++r) st[kt][r] = 0.f;
            if constexpr (FP8QK) {
                // MX fp8 QK^T: K=64 dims/instruction -> 4x fewer MFMAs
#pragma unroll
                for (int m = 0; m < NMX; ++m) {
                    const unsigned char* base =
                        &k8_lds[buf * (KVBLK * K8ROW) +
                                (kt * 32 + l32) * K8ROW + m * 64 + hi * 32];
                    i32x4 lo = *reinterpret_cast<const i32x4*>(base);
                    i32x4 hi4 = *reinterpret_cast<const i32x4*>(base + 16);
                    i32x8 af = {lo[0], lo[1], lo[2], lo[3],
                                hi4[0], hi4[1], hi4[2], hi4[3]};
                    st[kt] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
                        af, qf8[m], st[kt], 0, 0, 0, 127, 0, 127);
                }
            } else {
#pragma unroll
                for (int kk = 0; kk < KK; ++kk) {
                    bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
                        &k_lds[buf * (KVBLK * KPAD) + (kt * 32 + l32) * KPAD +
                               kk * 16 + hi * 8]);
                    st[kt] = mfma32x32x16(afrag, qfrag[kk], st[kt]);
                }
            }
        }
        if constexpr (!STATIC_PRIO) __builtin_amdgcn_s_setprio(0);
    };
    auto softmax_half = [&](int kv0, f32x16* st, bf16x8* pfrag) {
        float mx = -3e30f;
#pragma unroll
        for (int kt = 0; kt < 2; ++kt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int key = kv0 + kt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
                const float sv = (key < Sk) ? st[kt][r] : -3e30f;
                st[kt][r] = sv;
                mx = fmaxf(mx, sv);
            }
        mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
        const float mnew = fmaxf(m_run, mx * scale2);
        const float alpha = __builtin_amdgcn_exp2f(m_run - mnew);
        m_run = mnew;
        float ps = 0.f;
#pragma unroll
        for (int kt = 0; kt < 2; ++kt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const float pv_ =
                    __builtin_amdgcn_exp2f(fmaf(st[kt][r], scale2, -mnew));
                st[kt][r] = pv_;
                ps += pv_;
            }
        ps += __shfl_xor(ps, 32, 64);
        l_run = l_run * alpha + ps;
        if (alpha != 1.f) {
#pragma unroll
            for (int n = 0; n < NV; ++n)
#pragma unroll
                for (int r = 0; r < 16; ++r) o_acc[n][r] *= alpha;
        }
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
    };
    auto pv_half = [&](int buf, bf16x8* pfrag) {
        if constexpr (!STATIC_PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int c = 0; c < 4; ++c) {
#pragma unroll
            for (int n = 0; n < NV; ++n) {
                s16x4 alo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (LDS_P s16x4*)&v_lds[buf * (KVBLK * VROW) + trb +
                                         c * (16 * VROW) + n * 32]);
                s16x4 ahi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (LDS_P s16x4*)&v_lds[buf * (KVBLK * VROW) + trb +
                                         c * (16 * VROW) + 4 * VROW + n * 32]);
                bf16x8 va =
                    __builtin_shufflevector(alo, ahi, 0, 1, 2, 3, 4, 5, 6, 7);
                o_acc[n] = mfma32x32x16(va, pfrag[c], o_acc[n]);
            }
        }
        if constexpr (!STATIC_PRIO) __builtin_amdgcn_s_setprio(0);
    };

    const int n_tiles = (Sk + KVBLK - 1) / KVBLK;
    if constexpr (VAR == 4) {
        // 4-slot ring, barrier every SECOND tile. Writes land TWO tiles
        // ahead (tile t+2 from reg slot t%2) so every read is separated
        // from its write by at least one barrier; the 4-deep ring bounds
        // WAR hazards across the skew the sparser barriers allow.
        issue_tile_loads_slot(0, 0);
        write_k_slot(0, 0);
        write_v_slot(0, 0);
        if (n_tiles > 1) {
            issue_tile_loads_slot(KVBLK, 1);
            write_k_slot(1, 1);
            write_v_slot(1, 1);
        }
        if (n_tiles > 2) issue_tile_loads_slot(2 * KVBLK, 0);
        __syncthreads();
        for (int t = 0; t < n_tiles; ++t) {
            const int p = t & 3;
            f32x16 st[2];
            bf16x8 pfrag[4];
            qk_half(p, st);
            if (t + 2 < n_tiles) write_k_slot((t + 2) & 3, t & 1);
            softmax_half(t * KVBLK, st, pfrag);
            if (t + 2 < n_tiles) write_v_slot((t + 2) & 3, t & 1);
            if (t + 3 < n_tiles) issue_tile_loads_slot((t + 3) * KVBLK,
                                                       (t + 1) & 1);
            pv_half(p, pfrag);
            if (t & 1) __syncthreads();
        }
    } else if constexpr (VAR == 5) {
        // cross-tile stagger: [QK^T(t) ; PV(t-1)] in one barrier segment
        // (matrix beside matrix), [softmax(t) ; stores ; loads] in the
        // other. o_acc rescale order stays correct because softmax(t)
        // runs AFTER PV(t-1) lands its m(t-1)-unit contributions.
        issue_tile_loads(0);
        write_k_lds(0);
        write_v_lds(0);
        if (n_tiles > 1) issue_tile_loads(KVBLK);
        __syncthreads();
        bf16x8 pfrag[4];
        f32x16 st[2];
        for (int t = 0; t < n_tiles; ++t) {
            const int p = t & 1;
            qk_half(p, st);
            if (t > 0) pv_half(p ^ 1, pfrag);
            __syncthreads();  // v[p^1] free for the (t+1) stores below
            softmax_half(t * KVBLK, st, pfrag);
            if (t + 1 < n_tiles) {
                write_k_lds(p ^ 1);
                write_v_lds(p ^ 1);
            }
            if (t + 2 < n_tiles) issue_tile_loads((t + 2) * KVBLK);
            __syncthreads();  // k/v[p^1] visible before QK^T(t+1)
        }
        pv_half((n_tiles - 1) & 1, pfrag);  // drain the last tile's PV
    } else if (VAR == 0) {
        issue_tile_loads(0);
        for (int t = 0; t < n_tiles; ++t) {
            __syncthreads();
            write_k_lds(0);
            write_v_lds(0);
            __syncthreads();
            if (t + 1 < n_tiles) issue_tile_loads((t + 1) * KVBLK);
            f32x16 st[2];
            bf16x8 pfrag[4];
            qk_half(0, st);
            softmax_half(t * KVBLK, st, pfrag);
            pv_half(0, pfrag);
        }
    } else {
        // v5: double-buffered LDS, one barrier per tile. Prologue stages
        // tile 0 into buf 0 and leaves tile 1 in registers.
        issue_tile_loads(0);
        write_k_lds(0);
        write_v_lds(0);
        if (n_tiles > 1) issue_tile_loads(KVBLK);
        __syncthreads();
        for (int t = 0; t < n_tiles; ++t) {
            const int p = t & 1;
            f32x16 st[2];
            bf16x8 pfrag[4];
            if (VAR == 2) {
                // v5b: stores first (the whole stage ahead of compute)
                if (t + 1 < n_tiles) {
                    write_k_lds(p ^ 1);
                    write_v_lds(p ^ 1);
                }
                qk_half(p, st);
                softmax_half(t * KVBLK, st, pfrag);
                if (t + 2 < n_tiles) issue_tile_loads((t + 2) * KVBLK);
            } else {
                // v5a: stores interleaved where the LDS port is idle
                qk_half(p, st);
                if (t + 1 < n_tiles) write_k_lds(p ^ 1);
                softmax_half(t * KVBLK, st, pfrag);
                if (t + 1 < n_tiles) write_v_lds(p ^ 1);
                if (t + 2 < n_tiles) issue_tile_loads((t + 2) * KVBLK);
            }
            pv_half(p, pfrag);
            __syncthreads();
        }
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
                *reinterpret_cast<unsigned long long*>(
                    op + (long)row * ss + dim0) =
                    *reinterpret_cast<unsigned long long*>(pack);
            }
    }
}

#undef kreg
#undef vreg

// ---------------------------------------------------------------------------
// var8 — WAVE-SPECIALIZED schedule (MI355X_MICROARCH "tuned 8-wave bf16
// attention loop" regime): the two waves co-resident on each SIMD
// alternate roles per segment — one in a matrix-heavy compute segment,
// its partner staging the upcoming tile — because the per-SIMD matrix
// pipe is SERIALIZED between the two waves anyway (item 1), alternation
// costs no MFMA throughput while the load half's memory ops fully hide.
//
// Half A = waves 0-3 (q-rows 0..127 of the block), half B = waves 4-7
// (rows 128..255); waves w and w+4 share SIMD w. 4-slot LDS ring
// (151.6 KB -> 1 block/CU; the overlap now lives INSIDE the block).
// Segment sequence per tile t: [A computes t | B loads K(t+2)] barrier
// [B computes t | A loads V(t+2)] barrier.
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(512, 1) void attn_ws_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out,
    int S, int Sk, float scale, int H) {
    constexpr int KVBLK = 64;
    constexpr int WAVES = 8;
    constexpr int KPAD = D + 8;
    constexpr int VROW = 160;
    constexpr int KK = D / 16;
    constexpr int NV = D / 32;
    constexpr int RING = 4;
    constexpr int HTHREADS = 256;                       // threads per half
    constexpr int KVECS = (KVBLK * D) / (8 * HTHREADS); // 4 at D=128

    __shared__ bf16 k_lds[RING * KVBLK * KPAD];
    __shared__ bf16 v_lds[RING * KVBLK * VROW];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int half = wid >> 2;          // 0 = A, 1 = B
    const int hwid = wid & 3;           // wave id within the half
    const int htid = (hwid << 6) | lane;  // thread id within the half
    const int l32 = lane & 31;
    const int hi = lane >> 5;
    const int trb = (hi * 8 + ((lane & 15) >> 2)) * VROW +
                    16 * (((lane >> 4) & 1) ^ hi) + 4 * (lane & 3);

    const int nq = (S + WAVES * 32 - 1) / (WAVES * 32);
    const long id = blockIdx.x;
    const long bh = (id & 7) + 8 * ((id >> 3) / nq);
    const int qtile = (int)((id >> 3) % nq);
    const long b = bh / H;
    const int h = (int)(bh % H);
    // half A owns rows [0,128), half B rows [128,256) of the block
    const int q0 = qtile * (WAVES * 32) + half * 128 + hwid * 32;
    const int ss = H * D;

    const bf16* qp = q + (b * (long)S) * ss + (long)h * D;
    const bf16* kp = k + (b * (long)Sk) * ss + (long)h * D;
    const bf16* vp = v + (b * (long)Sk) * ss + (long)h * D;
    bf16* op = out + (b * (long)S) * ss + (long)h * D;

    bf16x8 qfrag[KK];
    {
        const int row = q0 + l32;
        const int rr = row < S ? row : S - 1;
#pragma unroll
        for (int kk = 0; kk < KK; ++kk)
            qfrag[kk] = *reinterpret_cast<const bf16x8*>(
                qp + (long)rr * ss + kk * 16 + hi * 8);
    }

    f32x16 o_acc[NV];
#pragma unroll
    for (int n = 0; n < NV; ++n)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[n][r] = 0.f;
    float m_run = -1e30f, l_run = 0.f;
    const float scale2 = scale * PA_LOG2E;

    // half-cooperative stagers (256 threads each)
    auto load_k = [&](int t) {
        const int slot = t & (RING - 1);
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = htid + i * HTHREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            const int src = t * KVBLK + row;
            bf16x8 val = (src < Sk)
                ? *reinterpret_cast<const bf16x8*>(kp + (long)src * ss + col)
                : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
            *reinterpret_cast<bf16x8*>(
                &k_lds[slot * (KVBLK * KPAD) + row * KPAD + col]) = val;
        }
    };
    auto load_v = [&](int t) {
        const int slot = t & (RING - 1);
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = htid + i * HTHREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            const int src = t * KVBLK + row;
            bf16x8 val = (src < Sk)
                ? *reinterpret_cast<const bf16x8*>(vp + (long)src * ss + col)
                : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
            *reinterpret_cast<bf16x8*>(
                &v_lds[slot * (KVBLK * VROW) + row * VROW +
                       (col ^ ((row & 8) << 1))]) = val;
        }
    };
    auto compute_tile = [&](int t) {
        const int buf = t & (RING - 1);
        const int kv0 = t * KVBLK;
        f32x16 st[2];
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
            for (int r = 0; r < 16; ++r) st[kt][r] = 0.f;
#pragma unroll
            for (int kk = 0; kk < KK; ++kk) {
                bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
                    &k_lds[buf * (KVBLK * KPAD) + (kt * 32 + l32) * KPAD +
                           kk * 16 + hi * 8]);
                st[kt] = mfma32x32x16(afrag, qfrag[kk], st[kt]);
            }
        }
        __builtin_amdgcn_s_setprio(0);
        float mx = -3e30f;
#pragma unroll
        for (int kt = 0; kt < 2; ++kt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int key = kv0 + kt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
                const float sv = (key < Sk) ? st[kt][r] : -3e30f;
                st[kt][r] = sv;
                mx = fmaxf(mx, sv);
            }
        mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
        const float mnew = fmaxf(m_run, mx * scale2);
        const float alpha = __builtin_amdgcn_exp2f(m_run - mnew);
        m_run = mnew;
        float ps = 0.f;
#pragma unroll
        for (int kt = 0; kt < 2; ++kt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const float pv_ =
                    __builtin_amdgcn_exp2f(fmaf(st[kt][r], scale2, -mnew));
                st[kt][r] = pv_;
                ps += pv_;
            }
        ps += __shfl_xor(ps, 32, 64);
        l_run = l_run * alpha + ps;
        if (alpha != 1.f) {
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
            unsigned int dd[4] = {(unsigned int)r02[0], (unsigned int)r13[0],
                                  (unsigned int)r02[1], (unsigned int)r13[1]};
            pfrag[c] = *reinterpret_cast<bf16x8*>(dd);
        }
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int c = 0; c < 4; ++c) {
#pragma unroll
            for (int n = 0; n < NV; ++n) {
                s16x4 alo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (LDS_P s16x4*)&v_lds[buf * (KVBLK * VROW) + trb +
                                         c * (16 * VROW) + n * 32]);
                s16x4 ahi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (LDS_P s16x4*)&v_lds[buf * (KVBLK * VROW) + trb +
                                         c * (16 * VROW) + 4 * VROW + n * 32]);
                bf16x8 va =
                    __builtin_shufflevector(alo, ahi, 0, 1, 2, 3, 4, 5, 6, 7);
                o_acc[n] = mfma32x32x16(va, pfrag[c], o_acc[n]);
            }
        }
        __builtin_amdgcn_s_setprio(0);
    };

    const int n_tiles = (Sk + KVBLK - 1) / KVBLK;
    // prologue: A stages tile 0 fully, B stages tile 1 fully
    if (half == 0) {
        load_k(0);
        load_v(0);
    } else if (n_tiles > 1) {
        load_k(1);
        load_v(1);
    }
    __syncthreads();
    for (int t = 0; t < n_tiles; ++t) {
        // segment 2t: A computes t, B stages K(t+2)
        if (half == 0) compute_tile(t);
        else if (t + 2 < n_tiles) load_k(t + 2);
        __syncthreads();
        // segment 2t+1: B computes t, A stages V(t+2)
        if (half == 1) compute_tile(t);
        else if (t + 2 < n_tiles) load_v(t + 2);
        __syncthreads();
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
                *reinterpret_cast<unsigned long long*>(
                    op + (long)row * ss + dim0) =
                    *reinterpret_cast<unsigned long long*>(pack);
            }
    }
}

// ---------------------------------------------------------------------------
// var9 — HALF-STAGGERED schedule: the two block halves run the SAME
// per-tile phase pair {P1 = QK^T+softmax, P2 = PV+stage} but offset by
// one phase, so on every SIMD one wave's softmax VALU issues beside its
// partner's PV MFMAs and vice versa (var8 showed an idle partner
// uncovers softmax; var5 showed both phases in ONE wave serialize).
// 2 barriers/tile, 4-slot ring (151.6 KB, 1 block/CU).
//   seg X: A: P1(t)              B: P2(t) + stage K(t+2)
//   seg Y: A: P2(t) + stage V(t+2)   B: P1(t+1)
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(512, 1) void attn_st_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out,
    int S, int Sk, float scale, int H) {
    constexpr int KVBLK = 64;
    constexpr int WAVES = 8;
    constexpr int KPAD = D + 8;
    constexpr int VROW = 160;
    constexpr int KK = D / 16;
    constexpr int NV = D / 32;
    constexpr int RING = 4;
    constexpr int HTHREADS = 256;
    constexpr int KVECS = (KVBLK * D) / (8 * HTHREADS);

    __shared__ bf16 k_lds[RING * KVBLK * KPAD];
    __shared__ bf16 v_lds[RING * KVBLK * VROW];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int half = wid >> 2;
    const int hwid = wid & 3;
    const int htid = (hwid << 6) | lane;
    const int l32 = lane & 31;
    const int hi = lane >> 5;
    const int trb = (hi * 8 + ((lane & 15) >> 2)) * VROW +
                    16 * (((lane >> 4) & 1) ^ hi) + 4 * (lane & 3);

    const int nq = (S + WAVES * 32 - 1) / (WAVES * 32);
    const long id = blockIdx.x;
    const long bh = (id & 7) + 8 * ((id >> 3) / nq);
    const int qtile = (int)((id >> 3) % nq);
    const long b = bh / H;
    const int h = (int)(bh % H);
    const int q0 = qtile * (WAVES * 32) + half * 128 + hwid * 32;
    const int ss = H * D;

    const bf16* qp = q + (b * (long)S) * ss + (long)h * D;
    const bf16* kp = k + (b * (long)Sk) * ss + (long)h * D;
    const bf16* vp = v + (b * (long)Sk) * ss + (long)h * D;
    bf16* op = out + (b * (long)S) * ss + (long)h * D;

    bf16x8 qfrag[KK];
    {
        const int row = q0 + l32;
        const int rr = row < S ? row : S - 1;
#pragma unroll
        for (int kk = 0; kk < KK; ++kk)
            qfrag[kk] = *reinterpret_cast<const bf16x8*>(
                qp + (long)rr * ss + kk * 16 + hi * 8);
    }

    f32x16 o_acc[NV];
#pragma unroll
    for (int n = 0; n < NV; ++n)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[n][r] = 0.f;
    float m_run = -1e30f, l_run = 0.f;
    const float scale2 = scale * PA_LOG2E;
    bf16x8 pfrag[4];  // carried from P1 to P2

    auto load_k = [&](int t) {
        const int slot = t & (RING - 1);
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = htid + i * HTHREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            const int src = t * KVBLK + row;
            bf16x8 val = (src < Sk)
                ? *reinterpret_cast<const bf16x8*>(kp + (long)src * ss + col)
                : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
            *reinterpret_cast<bf16x8*>(
                &k_lds[slot * (KVBLK * KPAD) + row * KPAD + col]) = val;
        }
    };
    auto load_v = [&](int t) {
        const int slot = t & (RING - 1);
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = htid + i * HTHREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            const int src = t * KVBLK + row;
            bf16x8 val = (src < Sk)
                ? *reinterpret_cast<const bf16x8*>(vp + (long)src * ss + col)
                : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
            *reinterpret_cast<bf16x8*>(
                &v_lds[slot * (KVBLK * VROW) + row * VROW +
                       (col ^ ((row & 8) << 1))]) = val;
        }
    };
    auto p1 = [&](int t) {  // QK^T + softmax + P fragments
        const int buf = t & (RING - 1);
        const int kv0 = t * KVBLK;
        f32x16 st[2];
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
            for (int r = 0; r < 16; ++r) st[kt][r] = 0.f;
#pragma unroll
            for (int kk = 0; kk < KK; ++kk) {
                bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
                    &k_lds[buf * (KVBLK * KPAD) + (kt * 32 + l32) * KPAD +
                           kk * 16 + hi * 8]);
                st[kt] = mfma32x32x16(afrag, qfrag[kk], st[kt]);
            }
        }
        __builtin_amdgcn_s_setprio(0);
        float mx = -3e30f;
#pragma unroll
        for (int kt = 0; kt < 2; ++kt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int key = kv0 + kt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
                const float sv = (key < Sk) ? st[kt][r] : -3e30f;
                st[kt][r] = sv;
                mx = fmaxf(mx, sv);
            }
        mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
        const float mnew = fmaxf(m_run, mx * scale2);
        const float alpha = __builtin_amdgcn_exp2f(m_run - mnew);
        m_run = mnew;
        float ps = 0.f;
#pragma unroll
        for (int kt = 0; kt < 2; ++kt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const float pv_ =
                    __builtin_amdgcn_exp2f(fmaf(st[kt][r], scale2, -mnew));
                st[kt][r] = pv_;
                ps += pv_;
            }
        ps += __shfl_xor(ps, 32, 64);
        l_run = l_run * alpha + ps;
        if (alpha != 1.f) {
#pragma unroll
            for (int n = 0; n < NV; ++n)
#pragma unroll
                for (int r = 0; r < 16; ++r) o_acc[n][r] *= alpha;
        }
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
            unsigned int dd[4] = {(unsigned int)r02[0], (unsigned int)r13[0],
                                  (unsigned int)r02[1], (unsigned int)r13[1]};
            pfrag[c] = *reinterpret_cast<bf16x8*>(dd);
        }
    };
    auto p2 = [&](int t) {  // PV from the carried pfrag
        const int buf = t & (RING - 1);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int c = 0; c < 4; ++c) {
#pragma unroll
            for (int n = 0; n < NV; ++n) {
                s16x4 alo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (LDS_P s16x4*)&v_lds[buf * (KVBLK * VROW) + trb +
                                         c * (16 * VROW) + n * 32]);
                s16x4 ahi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (LDS_P s16x4*)&v_lds[buf * (KVBLK * VROW) + trb +
                                         c * (16 * VROW) + 4 * VROW + n * 32]);
                bf16x8 va =
                    __builtin_shufflevector(alo, ahi, 0, 1, 2, 3, 4, 5, 6, 7);
                o_acc[n] = mfma32x32x16(va, pfrag[c], o_acc[n]);
            }
        }
        __builtin_amdgcn_s_setprio(0);
    };

    const int n_tiles = (Sk + KVBLK - 1) / KVBLK;
    // prologue: A stages tile 0, B stages tile 1; then B runs P1(0) while
    // A stages... B needs K0 (staged by A) -> barrier first.
    if (half == 0) {
        load_k(0);
        load_v(0);
    } else if (n_tiles > 1) {
        load_k(1);
        load_v(1);
    }
    __syncthreads();
    // offset segment: B leads with P1(0); A stages K(2)
    if (half == 1) p1(0);
    else if (n_tiles > 2) load_k(2);
    __syncthreads();
    for (int t = 0; t < n_tiles; ++t) {
        // seg X: A P1(t) | B P2(t) + stage V(t+2)
        if (half == 0) p1(t);
        else {
            p2(t);
            if (t + 2 < n_tiles) load_v(t + 2);
        }
        __syncthreads();
        // seg Y: A P2(t) + stage K(t+3) | B P1(t+1)
        if (half == 0) {
            p2(t);
            if (t + 3 < n_tiles) load_k(t + 3);
        } else if (t + 1 < n_tiles) {
            p1(t + 1);
        }
        __syncthreads();
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
                *reinterpret_cast<unsigned long long*>(
                    op + (long)row * ss + dim0) =
                    *reinterpret_cast<unsigned long long*>(pack);
            }
    }
}

// ---------------------------------------------------------------------------
// var6 — D=64 two-q-tile wave (NOTES r02 item 7): each wave owns 64 q-rows
// (two 32-row frames). K A-fragments and V tr-reads are shared between the
// two frames, halving LDS traffic per output row and amortizing the
// per-tile barrier/stage costs that double D=64's softmax fraction.
// v5a schedule (double-buffered LDS, one barrier per tile).
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(512, 2) void attn_d64x2_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out,
    int S, int Sk, float scale, int H) {
    constexpr int KVBLK = 64;
    constexpr int WAVES = 8;
    constexpr int THREADS = WAVES * 64;
    constexpr int KPAD = D + 8;
    constexpr int VROW = 160;
    constexpr int KK = D / 16;
    constexpr int NV = D / 32;
    constexpr int KVECS = (KVBLK * D) / (8 * THREADS);
    constexpr int QT = 2;

    __shared__ bf16 k_lds[2 * KVBLK * KPAD];
    __shared__ bf16 v_lds[2 * KVBLK * VROW];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l32 = lane & 31;
    const int hi = lane >> 5;
    const int trb = (hi * 8 + ((lane & 15) >> 2)) * VROW +
                    16 * (((lane >> 4) & 1) ^ hi) + 4 * (lane & 3);

    const int rows_per_blk = WAVES * 32 * QT;  // 512
    const int nq = (S + rows_per_blk - 1) / rows_per_blk;
    const long id = blockIdx.x;
    const long bh = (id & 7) + 8 * ((id >> 3) / nq);
    const int qtile = (int)((id >> 3) % nq);
    const long b = bh / H;
    const int h = (int)(bh % H);
    const int q0 = qtile * rows_per_blk + wid * (32 * QT);
    const int ss = H * D;

    const bf16* qp = q + (b * (long)S) * ss + (long)h * D;
    const bf16* kp = k + (b * (long)Sk) * ss + (long)h * D;
    const bf16* vp = v + (b * (long)Sk) * ss + (long)h * D;
    bf16* op = out + (b * (long)S) * ss + (long)h * D;

    bf16x8 qfrag[QT][KK];
#pragma unroll
    for (int qt = 0; qt < QT; ++qt) {
        const int row = q0 + qt * 32 + l32;
        const int rr = row < S ? row : S - 1;
#pragma unroll
        for (int kk = 0; kk < KK; ++kk)
            qfrag[qt][kk] = *reinterpret_cast<const bf16x8*>(
                qp + (long)rr * ss + kk * 16 + hi * 8);
    }

    f32x16 o_acc[QT][NV];
#pragma unroll
    for (int qt = 0; qt < QT; ++qt)
#pragma unroll
        for (int n = 0; n < NV; ++n)
#pragma unroll
            for (int r = 0; r < 16; ++r) o_acc[qt][n][r] = 0.f;
    float m_run[QT] = {-1e30f, -1e30f}, l_run[QT] = {0.f, 0.f};
    const float scale2 = scale * PA_LOG2E;

    bf16x8 kreg[KVECS], vreg[KVECS];
    auto issue_loads = [&](int kv0) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = tid + i * THREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            const int src = kv0 + row;
            if (src < Sk) {
                kreg[i] = *reinterpret_cast<const bf16x8*>(
                    kp + (long)src * ss + col);
                vreg[i] = *reinterpret_cast<const bf16x8*>(
                    vp + (long)src * ss + col);
            } else {
                kreg[i] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
                vreg[i] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
            }
        }
    };
    auto write_k = [&](int buf) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = tid + i * THREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            *reinterpret_cast<bf16x8*>(
                &k_lds[buf * (KVBLK * KPAD) + row * KPAD + col]) = kreg[i];
        }
    };
    auto write_v = [&](int buf) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i) {
            const int idx = tid + i * THREADS;
            const int row = idx / (D / 8);
            const int col = (idx % (D / 8)) * 8;
            *reinterpret_cast<bf16x8*>(
                &v_lds[buf * (KVBLK * VROW) + row * VROW +
                       (col ^ ((row & 8) << 1))]) = vreg[i];
        }
    };

    const int n_tiles = (Sk + KVBLK - 1) / KVBLK;
    issue_loads(0);
    write_k(0);
    write_v(0);
    if (n_tiles > 1) issue_loads(KVBLK);
    __syncthreads();
    for (int t = 0; t < n_tiles; ++t) {
        const int p = t & 1;
        const int kv0 = t * KVBLK;
        // QK^T both frames: the K A-fragment is read from LDS ONCE per
        // (kt, kk) and feeds both frames' MFMAs
        f32x16 st[QT][2];
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
            for (int qt = 0; qt < QT; ++qt)
#pragma unroll
                for (int r = 0; r < 16; ++r) st[qt][kt][r] = 0.f;
#pragma unroll
            for (int kk = 0; kk < KK; ++kk) {
                bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
                    &k_lds[p * (KVBLK * KPAD) + (kt * 32 + l32) * KPAD +
                           kk * 16 + hi * 8]);
#pragma unroll
                for (int qt = 0; qt < QT; ++qt)
                    st[qt][kt] = mfma32x32x16(afrag, qfrag[qt][kk],
                                              st[qt][kt]);
            }
        }
        __builtin_amdgcn_s_setprio(0);
        if (t + 1 < n_tiles) write_k(p ^ 1);

        // softmax + P per frame
        bf16x8 pfrag[QT][4];
#pragma unroll
        for (int qt = 0; qt < QT; ++qt) {
            float mx = -3e30f;
#pragma unroll
            for (int kt = 0; kt < 2; ++kt)
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int key =
                        kv0 + kt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
                    const float sv = (key < Sk) ? st[qt][kt][r] : -3e30f;
                    st[qt][kt][r] = sv;
                    mx = fmaxf(mx, sv);
                }
            mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
            const float mnew = fmaxf(m_run[qt], mx * scale2);
            const float alpha = __builtin_amdgcn_exp2f(m_run[qt] - mnew);
            m_run[qt] = mnew;
            float ps = 0.f;
#pragma unroll
            for (int kt = 0; kt < 2; ++kt)
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const float pv_ = __builtin_amdgcn_exp2f(
                        fmaf(st[qt][kt][r], scale2, -mnew));
                    st[qt][kt][r] = pv_;
                    ps += pv_;
                }
            ps += __shfl_xor(ps, 32, 64);
            l_run[qt] = l_run[qt] * alpha + ps;
            if (alpha != 1.f) {
#pragma unroll
                for (int n = 0; n < NV; ++n)
#pragma unroll
                    for (int r = 0; r < 16; ++r) o_acc[qt][n][r] *= alpha;
            }
#pragma unroll
            for (int c = 0; c < 4; ++c) {
                const f32x16& sv = st[qt][c >> 1];
                const int rb = 8 * (c & 1);
                unsigned int w0 = cvt_pk_bf16(sv[rb + 0], sv[rb + 1]);
                unsigned int w1 = cvt_pk_bf16(sv[rb + 2], sv[rb + 3]);
                unsigned int w2 = cvt_pk_bf16(sv[rb + 4], sv[rb + 5]);
                unsigned int w3 = cvt_pk_bf16(sv[rb + 6], sv[rb + 7]);
                auto r02 =
                    __builtin_amdgcn_permlane32_swap(w0, w2, false, false);
                auto r13 =
                    __builtin_amdgcn_permlane32_swap(w1, w3, false, false);
                unsigned int dd[4] = {(unsigned int)r02[0],
                                      (unsigned int)r13[0],
                                      (unsigned int)r02[1],
                                      (unsigned int)r13[1]};
                pfrag[qt][c] = *reinterpret_cast<bf16x8*>(dd);
            }
        }
        if (t + 1 < n_tiles) write_v(p ^ 1);
        if (t + 2 < n_tiles) issue_loads((t + 2) * KVBLK);

        // PV both frames: each V tr-read pair feeds both frames' MFMAs
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int c = 0; c < 4; ++c) {
#pragma unroll
            for (int n = 0; n < NV; ++n) {
                s16x4 alo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (LDS_P s16x4*)&v_lds[p * (KVBLK * VROW) + trb +
                                         c * (16 * VROW) + n * 32]);
                s16x4 ahi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                    (LDS_P s16x4*)&v_lds[p * (KVBLK * VROW) + trb +
                                         c * (16 * VROW) + 4 * VROW +
                                         n * 32]);
                bf16x8 va =
                    __builtin_shufflevector(alo, ahi, 0, 1, 2, 3, 4, 5, 6, 7);
#pragma unroll
                for (int qt = 0; qt < QT; ++qt)
                    o_acc[qt][n] = mfma32x32x16(va, pfrag[qt][c],
                                                o_acc[qt][n]);
            }
        }
        __builtin_amdgcn_s_setprio(0);
        __syncthreads();
    }

#pragma unroll
    for (int qt = 0; qt < QT; ++qt) {
        const int row = q0 + qt * 32 + l32;
        if (row < S) {
            const float inv_l = (l_run[qt] > 0.f) ? 1.f / l_run[qt] : 0.f;
#pragma unroll
            for (int n = 0; n < NV; ++n)
#pragma unroll
                for (int r2 = 0; r2 < 4; ++r2) {
                    const int dim0 = n * 32 + 8 * r2 + 4 * hi;
                    unsigned short pack[4];
#pragma unroll
                    for (int j = 0; j < 4; ++j)
                        pack[j] = __bfloat16_as_ushort(
                            f2bf(o_acc[qt][n][r2 * 4 + j] * inv_l));
                    *reinterpret_cast<unsigned long long*>(
                        op + (long)row * ss + dim0) =
                        *reinterpret_cast<unsigned long long*>(pack);
                }
        }
    }
}

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP error %d at %d\n", e, __LINE__); exit(1); } } while (0)

static void fill_random(bf16* dst, long n, unsigned seed) {
    std::vector<unsigned short> h(1 << 20);
    unsigned x = seed;
    for (auto& e : h) {
        x = x * 1664525u + 1013904223u;
        float f = ((x >> 8) / 8388608.0f) * 2.f - 1.f;
        unsigned int bits;
        __builtin_memcpy(&bits, &f, 4);
        e = (unsigned short)(bits >> 16);
    }
    for (long off = 0; off < n; off += (1 << 20)) {
        long len = std::min<long>(1 << 20, n - off);
        HIP_CHECK(hipMemcpy(dst + off, h.data(), len * 2,
                            hipMemcpyHostToDevice));
    }
}

// joint q/k fp8 scale: amax/448, set per shape by the host (the real
// integration computes it from the tensors; the spike test needs it too)
static float QK_S = 1.0f / 448.0f;

template <int D>
static void launch(int var, const bf16* q, const bf16* k, const bf16* v,
                   bf16* o, int B, int H, int S, float scale) {
    dim3 grid(((S + 255) / 256) * B * H), blk(512);
    const float inv_s = 1.0f / QK_S, s2 = QK_S * QK_S;
    switch (var) {
    case 0:
        hipLaunchKernelGGL((attn_v5_kernel<D, 0>), grid, blk, 0, 0, q, k, v,
                           o, S, S, scale, H, inv_s, s2);
        break;
    case 1:
        hipLaunchKernelGGL((attn_v5_kernel<D, 1>), grid, blk, 0, 0, q, k, v,
                           o, S, S, scale, H, inv_s, s2);
        break;
    case 2:
        hipLaunchKernelGGL((attn_v5_kernel<D, 2>), grid, blk, 0, 0, q, k, v,
                           o, S, S, scale, H, inv_s, s2);
        break;
    case 3:
        hipLaunchKernelGGL((attn_v5_kernel<D, 3>), grid, blk, 0, 0, q, k, v,
                           o, S, S, scale, H, inv_s, s2);
        break;
    case 4:
        hipLaunchKernelGGL((attn_v5_kernel<D, 4>), grid, blk, 0, 0, q, k, v,
                           o, S, S, scale, H, inv_s, s2);
        break;
    case 5:
        hipLaunchKernelGGL((attn_v5_kernel<D, 5>), grid, blk, 0, 0, q, k, v,
                           o, S, S, scale, H, inv_s, s2);
        break;
    case 6:
        // var6: two-q-tile wave — a D=64 remedy; at D=128 time v5a so the
        // interleaved rounds stay aligned
        if (D == 64) {
            dim3 g2(((S + 511) / 512) * B * H);
            hipLaunchKernelGGL((attn_d64x2_kernel<64>), g2, blk, 0, 0, q, k,
                               v, o, S, S, scale, H);
        } else {
            hipLaunchKernelGGL((attn_v5_kernel<D, 1>), grid, blk, 0, 0, q, k,
                               v, o, S, S, scale, H, inv_s, s2);
        }
        break;
    case 7:
        hipLaunchKernelGGL((attn_v5_kernel<D, 7>), grid, blk, 0, 0, q, k, v,
                           o, S, S, scale, H, inv_s, s2);
        break;
    case 8:
        // var8: wave-specialized (D=128 only; D=64 -> v5a control)
        if (D == 128)
            hipLaunchKernelGGL((attn_ws_kernel<128>), grid, blk, 0, 0, q, k,
                               v, o, S, S, scale, H);
        else
            hipLaunchKernelGGL((attn_v5_kernel<D, 1>), grid, blk, 0, 0, q, k,
                               v, o, S, S, scale, H, inv_s, s2);
        break;
    default:
        // var9: half-staggered phases (D=128 only; D=64 -> v5a control)
        if (D == 128)
            hipLaunchKernelGGL((attn_st_kernel<128>), grid, blk, 0, 0, q, k,
                               v, o, S, S, scale, H);
        else
            hipLaunchKernelGGL((attn_v5_kernel<D, 1>), grid, blk, 0, 0, q, k,
                               v, o, S, S, scale, H, inv_s, s2);
    }
}

constexpr int NVAR = 10;

template <int D>
static int check_correct(int B, int H, int S, bool spike) {
    const long n = (long)B * S * H * D;
    bf16 *q, *k, *v, *o;
    HIP_CHECK(hipMalloc(&q, n * 2));
    HIP_CHECK(hipMalloc(&k, n * 2));
    HIP_CHECK(hipMalloc(&v, n * 2));
    HIP_CHECK(hipMalloc(&o, n * 2));
    fill_random(q, n, 1234);
    fill_random(k, n, 777);
    fill_random(v, n, 4242);
    if (spike) {
        // spike a LATE K row so the running max jumps past tile 0's
        // (rule 26: force the rescale branch)
        std::vector<unsigned short> big(H * D, 0x4120);  // 10.0 bf16
        HIP_CHECK(hipMemcpy(k + (long)(S - 7) * H * D, big.data(),
                            big.size() * 2, hipMemcpyHostToDevice));
    }
    QK_S = (spike ? 10.0f : 1.0f) / 448.0f;  // true joint q/k amax
    const float scale = 1.0f / sqrtf((float)D);
    std::vector<unsigned short> ref(n), got(n), got2(n);
    launch<D>(0, q, k, v, o, B, H, S, scale);
    HIP_CHECK(hipGetLastError());
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(ref.data(), o, n * 2, hipMemcpyDeviceToHost));
    int fails = 0;
    for (int var = 1; var < NVAR; ++var) {
        HIP_CHECK(hipMemset(o, 0, n * 2));
        launch<D>(var, q, k, v, o, B, H, S, scale);
        HIP_CHECK(hipGetLastError());
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipMemcpy(got.data(), o, n * 2, hipMemcpyDeviceToHost));
        // race screen: second run must be bitwise identical
        HIP_CHECK(hipMemset(o, 0, n * 2));
        launch<D>(var, q, k, v, o, B, H, S, scale);
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipMemcpy(got2.data(), o, n * 2, hipMemcpyDeviceToHost));
        // var 3 quantizes q/k to e4m3 (~6% per-element) — compare with a
        // documented looser bound; bf16 variants stay at 5%.
        const float tol = (var == 3) ? 0.15f : 0.05f;
        long bad = 0, race = 0;
        double err_sum = 0.0, ref_sum = 0.0;
        for (long i = 0; i < n; ++i) {
            if (got[i] != got2[i]) ++race;
            float a, b2;
            unsigned int ua = (unsigned int)ref[i] << 16,
                         ub = (unsigned int)got[i] << 16;
            __builtin_memcpy(&a, &ua, 4);
            __builtin_memcpy(&b2, &ub, 4);
            err_sum += (a - b2) * (double)(a - b2);
            ref_sum += a * (double)a;
            if (fabsf(a - b2) > tol + tol * fabsf(a)) ++bad;
        }
        if (var == 3)
            printf("var3 fp8-qk rel-l2 error D=%d: %.4f\n", D,
                   sqrt(err_sum / (ref_sum + 1e-30)));
        if (bad || race) {
            printf("VAR %d WRONG D=%d S=%d spike=%d: %ld/%ld mismatch, "
                   "%ld race\n", var, D, S, (int)spike, bad, n, race);
            // var3's fp8 error is the quantity under study — report it
            // (above) but do not block the timing runs on it; any race
            // or a bf16-variant miss still fails hard.
            if (var != 3 || race) ++fails;
        }
    }
    (void)hipFree(q); (void)hipFree(k); (void)hipFree(v); (void)hipFree(o);
    return fails;
}

int main(int argc, char** argv) {
    const int rounds = argc > 1 ? atoi(argv[1]) : 8;
    // NOTE: B*H must be a multiple of 8 — the XCD-affine decode is only
    // bijective then (the shipped host wrapper gates on this; the harness
    // just uses conforming shapes). S=1000 exercises the 40-key tail tile.
    int fails = 0;
    fails += check_correct<128>(2, 4, 1000, false);
    fails += check_correct<128>(2, 4, 1000, true);
    fails += check_correct<64>(2, 4, 1000, false);
    fails += check_correct<64>(2, 4, 1000, true);
    if (fails) { printf("CORRECTNESS FAILED (%d)\n", fails); return 1; }
    printf("correctness: all variants agree (tails + spike + race screen)\n");

    struct Shape { int B, H, S, D; const char* name; };
    Shape shapes[3] = {{8, 24, 4608, 128, "flux"},
                       {1, 8, 30720, 128, "long"},
                       {8, 38, 4250, 64, "sd3-d64"}};
    for (auto& sh : shapes) {
        const int B = sh.B, H = sh.H, S = sh.S, D = sh.D;
        const long n = (long)B * S * H * D;
        bf16 *q, *k, *v, *o;
        HIP_CHECK(hipMalloc(&q, n * 2));
        HIP_CHECK(hipMalloc(&k, n * 2));
        HIP_CHECK(hipMalloc(&v, n * 2));
        HIP_CHECK(hipMalloc(&o, n * 2));
        fill_random(q, n, 12345);
        fill_random(k, n, 54321);
        fill_random(v, n, 999);
        const float scale = 1.0f / sqrtf((float)D);
        const double tf = 4.0 * B * H * (double)S * S * D / 1e12;
        double best[NVAR];
        for (auto& x : best) x = 1e30;
        auto run = [&](int var) {
            if (D == 64)
                launch<64>(var, q, k, v, o, B, H, S, scale);
            else
                launch<128>(var, q, k, v, o, B, H, S, scale);
        };
        run(0);
        HIP_CHECK(hipDeviceSynchronize());
        for (int r = 0; r < rounds; ++r)
            for (int var = 0; var < NVAR; ++var) {
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
        printf("%-8s", sh.name);
        for (int var = 0; var < NVAR; ++var)
            printf("  v%d %7.3f ms (%6.1f TF)", var, best[var],
                   tf / best[var] * 1e3);
        printf("\n");
        (void)hipFree(q); (void)hipFree(k); (void)hipFree(v); (void)hipFree(o);
    }
    return 0;
}
