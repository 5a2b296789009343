// Grouped expert GEMM for MoE (ref NxD blockwise-matmul MoE
// kernels via modules/moe_v2.py:104-161). One launch computes
//
//   y_sorted[m, n] = x_sorted[m, :] @ W[expert_of(m), n, :]^T
//
// over token rows SORTED by expert and padded per expert to MT-row
// tiles. Replaces the dense-bmm decode path's top_k/E redundant FLOPs
// (Mixtral batch 256: dense is compute-bound at ~2x the weight-stream
// floor) and the prefill gather loop's E x 3 launches + host syncs.
//
// Graph-safe contract: grid is STATIC for a given (T*top_k, N) — the
// host sizes it at the worst-case padded row count; blocks whose row
// tile has expert_id < 0 (beyond this step's real segments) write
// nothing. The per-step segment tables (expert id per row tile, sorted
// token order) are built with torch ops on device, so the whole MoE
// layer captures into one hipGraph.
//
// Structure = skinny_gemm5's verified pipeline (all-glds staging, 3-LDS
// buffer ring, ONE raw s_barrier per 64-deep K chunk, hand-counted
// vmcnt, XOR source-swizzle on both [rows][128B] images): the only
// deltas are the expert-indexed W base and the row-tile grid dim.

#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __bf16 mg_bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 mg_f16x8 __attribute__((ext_vector_type(8)));
typedef float mg_f32x4 __attribute__((ext_vector_type(4)));

template <typename T> struct MgMfma;
template <> struct MgMfma<__hip_bfloat16> {
    using vec8 = mg_bf16x8;
    static DEV_INLINE mg_f32x4 mfma(vec8 a, vec8 b, mg_f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    }
};
template <> struct MgMfma<__half> {
    using vec8 = mg_f16x8;
    static DEV_INLINE mg_f32x4 mfma(vec8 a, vec8 b, mg_f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
    }
};

constexpr int MG_BK = 64;
constexpr int MG_BN = 64;
constexpr int MG_NBUF = 3;

template <int AUX>
DEV_INLINE void mg_glds_rows128(const char* gsrc_rowbase, long grow_stride,
                                char* lds_dst_base, int maxrow = 7) {
    const int lane = threadIdx.x & 63;
    const int row = lane >> 3;
    const int slot = (lane & 7) ^ row;
    const int reff = max(0, min(row, maxrow));
    const char* src = gsrc_rowbase + (long)reff * grow_stride + slot * 16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)lds_dst_base, 16, 0, AUX);
}

template <int N>
DEV_INLINE void mg_vm_wait() {
    asm volatile("s_waitcnt vmcnt(%c0)" ::"i"(N) : "memory");
}

// per-lane-addressed glds (row gather): the caller precomputes each
// lane's full source address (row via an index table + XOR slot); the
// LDS dest stays the wave-uniform lane-linear base.
DEV_INLINE void mg_glds_lane128(const char* src_lane, char* lds_dst_base) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src_lane,
        (__attribute__((address_space(3))) unsigned int*)lds_dst_base, 16, 0, 0);
}

// MT=64 row tiles, BNT-wide column tiles (8 waves as 2(M) x 4(N): each
// wave = 32 rows x BNT/4 cols). Wider BNT cuts the x re-read traffic
// (x is re-read once per column tile) and raises MFMA work per staged
// byte; BNT=256 costs occupancy 1 (120 KB LDS) but each block then
// streams 4x the W per x fetch.
// BKT: k-elems per ring step (64 or 128). The pipeline is chunk-COUNT
// bound at these tile sizes (fixed wait/barrier cost per step), so the
// deeper step — staged as (BKT/64) 128B half-images per operand —
// halves the step count where LDS permits (bn<=128 at BKT=128).
// MTT: rows per tile. 32 halves the padded-row waste of fine-grained
// MoE (128 experts top-8 decode runs ~16 tokens/expert — a 64-row tile
// pads ~4x). At MTT=32 all 8 waves span the SAME 32 rows and split the
// columns 8 ways (BNT >= 128), and only waves 0..3 stage the four 1 KB
// x pieces (their vmcnt count is one higher — wave-uniform branch).
template <typename T, int BNT, int BKT = 64, int MTT = 64>
__global__ __launch_bounds__(512, 2) void moe_grouped_kernel(
    T* __restrict__ y,              // (Mpad, N) sorted-space output
    const T* __restrict__ x,        // (Mpad, K) sorted — or (T, K) with x_gather
    const T* __restrict__ w,        // (E, N, K) stacked expert weights
    const int* __restrict__ expert_ids,  // (n_row_tiles,) expert or -1
    const T* __restrict__ bias,     // (E, N) per-expert bias or null
    const int* __restrict__ x_gather,    // (Mpad,) padded row -> x row, or null
    float* __restrict__ combine_out,     // (T, N) fp32 accum, or null
    const float* __restrict__ combine_scale,  // (Mpad,) routing weight
    const int* __restrict__ combine_gather,   // (Mpad,) padded row -> token
    const int N, const int K) {
    using MF = MgMfma<T>;
    using vec8 = typename MF::vec8;
    constexpr int MT = MTT;
    static_assert(MTT == 64 || (MTT == 32 && BKT == 64 && BNT >= 128),
                  "MTT=32 supports BKT=64 and BNT>=128");
    constexpr int NH = BKT / 64;    // 128B half-images per operand per step
    constexpr int NWG = BNT / 64;   // W glds per wave per half-image
    constexpr int NXW = MTT / 8;    // waves that stage an x piece (4 or 8)
    constexpr int G = (MTT == 64 ? (1 + NWG) * NH : NWG);  // per ordinary wave
    constexpr int NFRAG = (MTT == 64 ? BNT / 64 : BNT / 128);
    constexpr int WNW = (MTT == 64 ? 4 : 8);   // waves along N
    constexpr int XBYTES = NH * MT * 128;
    constexpr int CHUNK_B = XBYTES + NH * BNT * 128;

    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tile = blockIdx.x;    // n tile
    const int rt = blockIdx.y;      // row tile
    const int e = expert_ids[rt];
    if (e < 0) return;              // beyond this step's padded rows
    const int n0 = tile * BNT;
    const int m0 = rt * MT;
    const int nchunks = K / BKT;

    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int wm0 = (MTT == 64 ? (wid >> 2) * 32 : 0);
    const int wn0 = (wid % WNW) * (BNT / WNW);
    const int lrow = lane & 15;
    const int lslot = lane >> 4;

    // per-lane x source: row via gather table (fused token gather) or
    // linear sorted row; XOR source-swizzle baked into the address.
    // MTT=32: only waves 0..3 stage x (one 8-row piece each)
    const bool x_wave = (MTT == 64) || wid < NXW;
    const int xr_lin = m0 + (x_wave ? wid * 8 : 0) + (lane >> 3);
    const int xr = x_gather ? max(x_gather[xr_lin], 0) : xr_lin;
    const int xslot = (lane & 7) ^ (lane >> 3);
    const char* xsrc_lane = reinterpret_cast<const char*>(x) +
                            (long)xr * K * sizeof(T) + xslot * 16;
    const char* wsrc[NWG];
#pragma unroll
    for (int i = 0; i < NWG; ++i)
        wsrc[i] = reinterpret_cast<const char*>(w) +
                  (((long)e * N + n0 + wid * (8 * NWG) + i * 8) * K) * sizeof(T);

    char* xdst[MG_NBUF];
    char* wdst[MG_NBUF];
#pragma unroll
    for (int b = 0; b < MG_NBUF; ++b) {
        xdst[b] = smem + b * CHUNK_B + (x_wave ? wid : 0) * 1024;
        wdst[b] = smem + b * CHUNK_B + XBYTES + wid * (NWG * 1024);
    }

#define MG_STAGE(t, b)                                                       \
    do {                                                                     \
        const long koff = (long)(t) * (BKT * 2);                             \
        _Pragma("unroll") for (int h = 0; h < NH; ++h) {                     \
            if (x_wave)                                                      \
                mg_glds_lane128(xsrc_lane + koff + h * 128,                  \
                                xdst[b] + h * (MT * 128));                   \
            _Pragma("unroll") for (int i = 0; i < NWG; ++i)                  \
                mg_glds_rows128<2>(wsrc[i] + koff + h * 128,                 \
                                   (long)K * sizeof(T),                      \
                                   wdst[b] + h * (BNT * 128) + i * 1024);    \
        }                                                                    \
    } while (0)

    mg_f32x4 acc[2][NFRAG];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
        for (int nj = 0; nj < NFRAG; ++nj) acc[qi][nj] = {0.f, 0.f, 0.f, 0.f};

#define MG_FRAGS(b)                                                          \
    do {                                                                     \
        _Pragma("unroll") for (int kq = 0; kq < 2 * NH; ++kq) {              \
            const char* xb =                                                 \
                smem + (b) * CHUNK_B + (kq >> 1) * (MT * 128);               \
            const char* wb =                                                 \
                smem + (b) * CHUNK_B + XBYTES + (kq >> 1) * (BNT * 128);     \
            const int ks = kq & 1;                                           \
            vec8 a0, a1, bb[NFRAG];                                          \
            {                                                                \
                const int m = wm0 + lrow;                                    \
                const int s = (ks * 4 + lslot) ^ (m & 7);                    \
                a0 = *reinterpret_cast<const vec8*>(xb + m * 128 + s * 16);  \
            }                                                                \
            {                                                                \
                const int m = wm0 + 16 + lrow;                               \
                const int s = (ks * 4 + lslot) ^ (m & 7);                    \
                a1 = *reinterpret_cast<const vec8*>(xb + m * 128 + s * 16);  \
            }                                                                \
            _Pragma("unroll") for (int nj = 0; nj < NFRAG; ++nj) {           \
                const int n = wn0 + nj * 16 + lrow;                          \
                const int s = (ks * 4 + lslot) ^ (n & 7);                    \
                bb[nj] = *reinterpret_cast<const vec8*>(wb + n * 128 + s * 16); \
            }                                                                \
            _Pragma("unroll") for (int nj = 0; nj < NFRAG; ++nj) {           \
                acc[0][nj] = MF::mfma(a0, bb[nj], acc[0][nj]);               \
                acc[1][nj] = MF::mfma(a1, bb[nj], acc[1][nj]);               \
            }                                                                \
        }                                                                    \
    } while (0)

#define MG_BODY(t, b, b2)                                                    \
    do {                                                                     \
        if ((t) + 1 < nchunks) {                                             \
            if (MTT == 32 && x_wave) mg_vm_wait<G + 1>();                    \
            else mg_vm_wait<G>();                                            \
        } else mg_vm_wait<0>();                                              \
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                   \
        __builtin_amdgcn_s_barrier();                                        \
        MG_FRAGS(b);                                                         \
        if ((t) + 2 < nchunks) MG_STAGE((t) + 2, b2);                        \
    } while (0)

    MG_STAGE(0, 0);
    if (nchunks > 1) MG_STAGE(1, 1);
    int t = 0;
    while (t + 3 <= nchunks) {
        MG_BODY(t + 0, 0, 2);
        MG_BODY(t + 1, 1, 0);
        MG_BODY(t + 2, 2, 1);
        t += 3;
    }
    if (t < nchunks) MG_BODY(t, 0, 2);
    if (t + 1 < nchunks) MG_BODY(t + 1, 1, 0);
#undef MG_BODY
#undef MG_FRAGS
#undef MG_STAGE
    mg_vm_wait<0>();

    // epilogue: y[m0 + wm0 + qi*16 + lslot*4 + r][n0 + wn0 + nj*16 + lrow]
    // — or, fused combine: atomically accumulate the routing-weighted
    // contribution straight into the (T, N) fp32 output (one rmw per
    // element, NOT per k-chunk — the v2 atomics lesson)
#pragma unroll
    for (int nj = 0; nj < NFRAG; ++nj) {
        const int n_out = n0 + wn0 + nj * 16 + lrow;
        const float bv =
            bias ? VecTraits<T>::to_float(bias[(long)e * N + n_out]) : 0.f;
#pragma unroll
        for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm0 + qi * 16 + lslot * 4 + r;
                if (combine_out) {
                    const int tk = combine_gather[m];
                    if (tk >= 0) {
                        const float sc =
                            combine_scale ? combine_scale[m] : 1.f;
                        atomicAdd(&combine_out[(long)tk * N + n_out],
                                  (acc[qi][nj][r] + bv) * sc);
                    }
                } else {
                    y[(long)m * N + n_out] =
                        VecTraits<T>::from_float(acc[qi][nj][r] + bv);
                }
            }
        }
    }
}


// ---------------------------------------------------------------------------
// MXFP4 variant (ref models/gpt_oss/mx_layout_transform.py):
// expert weights stay PACKED in HBM as OCP MXFP4 — e2m1 nibbles
// (E, N, K/2) u8 + e8m0 block scales (E, N, K/32) u8, 4.25 bits/weight —
// and are dequantized to the bf16 LDS image during staging (register
// stage + ds_write; x keeps glds). Weight traffic drops ~3.8x vs bf16,
// which is the decode-side win on this memory-bound path; the fp4-MFMA
// compute path (mfma_scale_*_f8f6f4) is a further step, not needed to
// beat the bf16 stream.

// decode 8 e2m1 nibbles (packed u32) * e8m0 scale into bf16x8.
// magnitude table x2 packed as nibbles: [0,1,2,3,4,6,8,12] (= 2*|v|)
template <typename T>
DEV_INLINE void mx4_dequant8(unsigned p, float half_scale, T* out) {
    constexpr unsigned MAG2 = 0xC8643210u;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        const unsigned nib = (p >> (4 * j)) & 0xF;
        float f = (float)((MAG2 >> ((nib & 7) * 4)) & 0xF);
        f = (nib & 8) ? -f : f;
        out[j] = VecTraits<T>::from_float(f * half_scale);
    }
}

template <typename T, int BNT>
__global__ __launch_bounds__(512, 2) void moe_grouped_mx4_kernel(
    T* __restrict__ y,                   // (Mpad, N)
    const T* __restrict__ x,             // (Mpad, K) — or (T, K) w/ x_gather
    const unsigned char* __restrict__ wq,  // (E, N, K/2) packed e2m1
    const unsigned char* __restrict__ ws,  // (E, N, K/32) e8m0 scales
    const int* __restrict__ expert_ids,  // (n_row_tiles,)
    const T* __restrict__ bias,          // (E, N) or null
    const int* __restrict__ x_gather,    // (Mpad,) or null
    float* __restrict__ combine_out,     // (T, N) fp32 or null
    const float* __restrict__ combine_scale,  // (Mpad,)
    const int* __restrict__ combine_gather,   // (Mpad,)
    const int N, const int K) {
    using MF = MgMfma<T>;
    using vec8 = typename MF::vec8;
    constexpr int MT = 64;
    constexpr int G = 1;                 // glds per wave per chunk (x only)
    constexpr int NWG = BNT / 64;        // W row groups per lane per chunk
    constexpr int NFRAG = BNT / 64;
    constexpr int XBYTES = MT * 128;
    constexpr int CHUNK_B = XBYTES + BNT * 128;

    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tile = blockIdx.x;
    const int rt = blockIdx.y;
    const int e = expert_ids[rt];
    if (e < 0) return;
    const int n0 = tile * BNT;
    const int m0 = rt * MT;
    const int nchunks = K / MG_BK;

    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int wm0 = (wid >> 2) * 32;
    const int wn0 = (wid & 3) * (BNT / 4);
    const int lrow = lane & 15;
    const int lslot = lane >> 4;

    const int xr_lin = m0 + wid * 8 + (lane >> 3);
    const int xr = x_gather ? max(x_gather[xr_lin], 0) : xr_lin;
    const int xslot = (lane & 7) ^ (lane >> 3);
    const char* xsrc_lane = reinterpret_cast<const char*>(x) +
                            (long)xr * K * sizeof(T) + xslot * 16;
    // this lane stages W rows (n0 + g*64 + wid*8 + lane/8), k-slot lane%8
    const int wrow_l = wid * 8 + (lane >> 3);
    const int wslot = lane & 7;          // 8 bf16 = 4 packed bytes
    const unsigned char* wq_row[NWG];
    const unsigned char* ws_row[NWG];
#pragma unroll
    for (int g = 0; g < NWG; ++g) {
        wq_row[g] = wq + ((long)e * N + n0 + g * 64 + wrow_l) * (K / 2);
        ws_row[g] = ws + ((long)e * N + n0 + g * 64 + wrow_l) * (K / 32);
    }

    char* xdst[MG_NBUF];
    char* wdst[MG_NBUF];
#pragma unroll
    for (int b = 0; b < MG_NBUF; ++b) {
        xdst[b] = smem + b * CHUNK_B + wid * 1024;
        wdst[b] = smem + b * CHUNK_B + XBYTES;
    }

    // x: glds with counted vmcnt; W: ordinary loads + in-register
    // dequant + swizzled ds_write (the ds_write slot XOR matches the
    // read-side XOR; note hipcc waits vmcnt(0) for the W loads while a
    // glds is in flight — the known mixed-kind cost, acceptable v1)
#define MX_STAGE_X(t, b)                                                     \
    mg_glds_lane128(xsrc_lane + (long)(t) * 128, xdst[b])
#define MX_STAGE_W(t, b)                                                     \
    do {                                                                     \
        const int kc = (t) * MG_BK;                                          \
        _Pragma("unroll") for (int g = 0; g < NWG; ++g) {                    \
            const unsigned pk = *reinterpret_cast<const unsigned*>(          \
                wq_row[g] + (kc + wslot * 8) / 2);                           \
            const float sc =                                                 \
                exp2f((int)ws_row[g][(kc + wslot * 8) / 32] - 127);          \
            T tmp[8];                                                        \
            mx4_dequant8<T>(pk, 0.5f * sc, tmp);                             \
            const int sl = wslot ^ (lane >> 3);                              \
            *reinterpret_cast<int4*>(wdst[b] + (g * 64 + wrow_l) * 128 +     \
                                     sl * 16) = *reinterpret_cast<int4*>(tmp); \
        }                                                                    \
    } while (0)

    mg_f32x4 acc[2][NFRAG];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
        for (int nj = 0; nj < NFRAG; ++nj) acc[qi][nj] = {0.f, 0.f, 0.f, 0.f};

#define MX_FRAGS(b)                                                          \
    do {                                                                     \
        const char* xb = smem + (b) * CHUNK_B;                               \
        const char* wb = xb + XBYTES;                                        \
        _Pragma("unroll") for (int ks = 0; ks < 2; ++ks) {                   \
            vec8 a0, a1, bb[NFRAG];                                          \
            {                                                                \
                const int m = wm0 + lrow;                                    \
                const int sfrag = (ks * 4 + lslot) ^ (m & 7);                \
                a0 = *reinterpret_cast<const vec8*>(xb + m * 128 + sfrag * 16); \
            }                                                                \
            {                                                                \
                const int m = wm0 + 16 + lrow;                               \
                const int sfrag = (ks * 4 + lslot) ^ (m & 7);                \
                a1 = *reinterpret_cast<const vec8*>(xb + m * 128 + sfrag * 16); \
            }                                                                \
            _Pragma("unroll") for (int nj = 0; nj < NFRAG; ++nj) {           \
                const int n = wn0 + nj * 16 + lrow;                          \
                const int sfrag = (ks * 4 + lslot) ^ (n & 7);                \
                bb[nj] = *reinterpret_cast<const vec8*>(wb + n * 128 + sfrag * 16); \
            }                                                                \
            _Pragma("unroll") for (int nj = 0; nj < NFRAG; ++nj) {           \
                acc[0][nj] = MF::mfma(a0, bb[nj], acc[0][nj]);               \
                acc[1][nj] = MF::mfma(a1, bb[nj], acc[1][nj]);               \
            }                                                                \
        }                                                                    \
    } while (0)

#define MX_BODY(t, b, b2)                                                    \
    do {                                                                     \
        if ((t) + 1 < nchunks) mg_vm_wait<G>();                              \
        else mg_vm_wait<0>();                                                \
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                   \
        __builtin_amdgcn_s_barrier();                                        \
        MX_FRAGS(b);                                                         \
        if ((t) + 2 < nchunks) {                                             \
            MX_STAGE_X((t) + 2, b2);                                         \
            MX_STAGE_W((t) + 2, b2);                                         \
        }                                                                    \
    } while (0)

    MX_STAGE_X(0, 0);
    MX_STAGE_W(0, 0);
    if (nchunks > 1) { MX_STAGE_X(1, 1); MX_STAGE_W(1, 1); }
    int t = 0;
    while (t + 3 <= nchunks) {
        MX_BODY(t + 0, 0, 2);
        MX_BODY(t + 1, 1, 0);
        MX_BODY(t + 2, 2, 1);
        t += 3;
    }
    if (t < nchunks) MX_BODY(t, 0, 2);
    if (t + 1 < nchunks) MX_BODY(t + 1, 1, 0);
#undef MX_BODY
#undef MX_FRAGS
#undef MX_STAGE_W
#undef MX_STAGE_X
    mg_vm_wait<0>();

#pragma unroll
    for (int nj = 0; nj < NFRAG; ++nj) {
        const int n_out = n0 + wn0 + nj * 16 + lrow;
        const float bv =
            bias ? VecTraits<T>::to_float(bias[(long)e * N + n_out]) : 0.f;
#pragma unroll
        for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm0 + qi * 16 + lslot * 4 + r;
                if (combine_out) {
                    const int tk = combine_gather[m];
                    if (tk >= 0) {
                        const float sc =
                            combine_scale ? combine_scale[m] : 1.f;
                        atomicAdd(&combine_out[(long)tk * N + n_out],
                                  (acc[qi][nj][r] + bv) * sc);
                    }
                } else {
                    y[(long)m * N + n_out] =
                        VecTraits<T>::from_float(acc[qi][nj][r] + bv);
                }
            }
        }
    }
}

// ---------------------------------------------------------------------------
// MXFP4 v3 ("mx4b"): all-glds packed-weight grouped GEMM for shapes with
// K % 256 == 0 (Mixtral/Llama-style MoE dims; GPT-OSS K=2880 keeps v1).
// A 256-deep K chunk packs to EXACTLY the [rows][128B] XOR-swizzled glds
// image of the bf16 kernel (256 e2m1 nibbles = 128 B), so the packed
// weights stream through the verified pipeline at full step depth —
// 3.8x less weight traffic. x stages as FOUR 128B half-images.
// The e8m0 block scales stage as a glds side-image too (16 B per row =
// the scales of chunks t and t+1; ~4% redundant, but it keeps EVERY
// load in the counted-glds pipeline — per-fragment global scale reads
// would re-trigger the mixed-load vmcnt drain that sank mx4 v1).
// Scale pieces are staged by waves 0..NS-1 only; s_waitcnt is wave-
// uniform, so those waves just use a count one higher.
// Dequant (nibble magnitude table * e8m0) happens at the LDS read.
// LDS at bn=128: 3 x (32K x + 16K W + 2K s) = 150 KB.

typedef float mg_f32x2 __attribute__((ext_vector_type(2)));

// gfx950 hardware e2m1 decode: one cvt_scalef32_pk_f32_fp4 converts 2
// nibbles AND multiplies by the f32 scale — ~4x fewer VALU than the
// nibble-table path (v1 keeps the table for portability of thought,
// not of code)
template <typename T>
DEV_INLINE void mx4b_hw_dequant8(unsigned p, float scale, T* out) {
    // sel must be an immediate
    mg_f32x2 v0 = __builtin_amdgcn_cvt_scalef32_pk_f32_fp4(p, scale, 0);
    mg_f32x2 v1 = __builtin_amdgcn_cvt_scalef32_pk_f32_fp4(p, scale, 1);
    mg_f32x2 v2 = __builtin_amdgcn_cvt_scalef32_pk_f32_fp4(p, scale, 2);
    mg_f32x2 v3 = __builtin_amdgcn_cvt_scalef32_pk_f32_fp4(p, scale, 3);
    out[0] = VecTraits<T>::from_float(v0[0]);
    out[1] = VecTraits<T>::from_float(v0[1]);
    out[2] = VecTraits<T>::from_float(v1[0]);
    out[3] = VecTraits<T>::from_float(v1[1]);
    out[4] = VecTraits<T>::from_float(v2[0]);
    out[5] = VecTraits<T>::from_float(v2[1]);
    out[6] = VecTraits<T>::from_float(v3[0]);
    out[7] = VecTraits<T>::from_float(v3[1]);
}

template <typename T, int BNT>
__global__ __launch_bounds__(512, 2) void moe_grouped_mx4b_kernel(
    T* __restrict__ y,                   // (Mpad, N)
    const T* __restrict__ x,             // (Mpad, K) — or (T, K) w/ x_gather
    const unsigned char* __restrict__ wq,  // (E, N, K/2) packed e2m1
    const unsigned char* __restrict__ ws,  // (E, N, K/32) e8m0 scales
    const int* __restrict__ expert_ids,  // (n_row_tiles,)
    const T* __restrict__ bias,          // (E, N) or null
    const int* __restrict__ x_gather,    // (Mpad,) or null
    float* __restrict__ combine_out,     // (T, N) fp32 or null
    const float* __restrict__ combine_scale,  // (Mpad,)
    const int* __restrict__ combine_gather,   // (Mpad,)
    const int N, const int K) {
    using MF = MgMfma<T>;
    using vec8 = typename MF::vec8;
    constexpr int MT = 64;
    constexpr int BKB = 256;        // k-elems per ring step
    constexpr int NH = 4;           // x 128B half-images per step
    constexpr int NWG = BNT / 64;   // W glds per wave per step
    constexpr int NS = BNT / 64;    // scale pieces per step (64 rows each)
    constexpr int G = NH + NWG;     // glds per ORDINARY wave per step
    constexpr int NFRAG = BNT / 64;
    constexpr int XBYTES = NH * MT * 128;
    constexpr int WBYTES = BNT * 128;
    constexpr int CHUNK_B = XBYTES + WBYTES + BNT * 16;
    static_assert(BNT == 64 || BNT == 128, "mx4b: bn 64/128 only");

    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tile = blockIdx.x;
    const int rt = blockIdx.y;
    const int e = expert_ids[rt];
    if (e < 0) return;
    const int n0 = tile * BNT;
    const int m0 = rt * MT;
    const int nchunks = K / BKB;
    const int srow_bytes = K / 32;  // scale bytes per weight row

    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int wm0 = (wid >> 2) * 32;
    const int wn0 = (wid & 3) * (BNT / 4);
    const int lrow = lane & 15;
    const int lslot = lane >> 4;

    const int xr_lin = m0 + wid * 8 + (lane >> 3);
    const int xr = x_gather ? max(x_gather[xr_lin], 0) : xr_lin;
    const int xslot = (lane & 7) ^ (lane >> 3);
    const char* xsrc_lane = reinterpret_cast<const char*>(x) +
                            (long)xr * K * sizeof(T) + xslot * 16;
    // packed W rows: K/2 bytes; 128 B per row per chunk -> standard
    // 8-rows-per-1KB pieces with the XOR source swizzle
    const char* wsrc[NWG];
#pragma unroll
    for (int i = 0; i < NWG; ++i)
        wsrc[i] = reinterpret_cast<const char*>(wq) +
                  ((long)e * N + n0 + wid * (8 * NWG) + i * 8) * (K / 2);
    // scale piece (waves 0..NS-1): lane covers ONE row's 16 B
    const bool sc_wave = wid < NS;
    const char* ssrc_row = reinterpret_cast<const char*>(ws) +
                           ((long)e * N + n0 + wid * 64 + lane) * srow_bytes;

    char* xdst[MG_NBUF];
    char* wdst[MG_NBUF];
    char* sdst[MG_NBUF];
#pragma unroll
    for (int b = 0; b < MG_NBUF; ++b) {
        xdst[b] = smem + b * CHUNK_B + wid * 1024;
        wdst[b] = smem + b * CHUNK_B + XBYTES + wid * (NWG * 1024);
        sdst[b] = smem + b * CHUNK_B + XBYTES + WBYTES + wid * 1024;
    }

#define B3_STAGE(t, b)                                                       \
    do {                                                                     \
        const long xoff = (long)(t) * 512;                                   \
        _Pragma("unroll") for (int h = 0; h < NH; ++h)                       \
            mg_glds_lane128(xsrc_lane + xoff + h * 128,                      \
                            xdst[b] + h * (MT * 128));                       \
        _Pragma("unroll") for (int i = 0; i < NWG; ++i)                      \
            mg_glds_rows128<2>(wsrc[i] + (long)(t) * 128, (long)(K / 2),     \
                               wdst[b] + i * 1024);                          \
        if (sc_wave)                                                         \
            mg_glds_lane128(ssrc_row + min((t) * 8, srow_bytes - 16),        \
                            sdst[b]);                                        \
    } while (0)

    mg_f32x4 acc[2][NFRAG];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
        for (int nj = 0; nj < NFRAG; ++nj) acc[qi][nj] = {0.f, 0.f, 0.f, 0.f};

#define B3_FRAGS(t, b)                                                       \
    do {                                                                     \
        const char* wb = smem + (b) * CHUNK_B + XBYTES;                      \
        const char* sb = smem + (b) * CHUNK_B + XBYTES + WBYTES;             \
        const int sdelta = (t) * 8 - min((t) * 8, srow_bytes - 16);          \
        _Pragma("unroll") for (int kq = 0; kq < 8; ++kq) {                   \
            const char* xh =                                                 \
                smem + (b) * CHUNK_B + (kq >> 1) * (MT * 128);               \
            const int ks = kq & 1;                                           \
            vec8 a0, a1, bb[NFRAG];                                          \
            {                                                                \
                const int m = wm0 + lrow;                                    \
                const int sx = (ks * 4 + lslot) ^ (m & 7);                   \
                a0 = *reinterpret_cast<const vec8*>(xh + m * 128 + sx * 16); \
            }                                                                \
            {                                                                \
                const int m = wm0 + 16 + lrow;                               \
                const int sx = (ks * 4 + lslot) ^ (m & 7);                   \
                a1 = *reinterpret_cast<const vec8*>(xh + m * 128 + sx * 16); \
            }                                                                \
            _Pragma("unroll") for (int nj = 0; nj < NFRAG; ++nj) {           \
                const int n = wn0 + nj * 16 + lrow;                          \
                const int s16 = kq ^ (n & 7);                                \
                const unsigned pk = *reinterpret_cast<const unsigned*>(      \
                    wb + n * 128 + s16 * 16 + lslot * 4);                    \
                const unsigned char se = *reinterpret_cast<                  \
                    const unsigned char*>(sb + n * 16 + sdelta + kq);        \
                const float sc = exp2f((int)se - 127);                       \
                T tmp[8];                                                    \
                mx4b_hw_dequant8<T>(pk, sc, tmp);                            \
                bb[nj] = *reinterpret_cast<const vec8*>(tmp);                \
            }                                                                \
            _Pragma("unroll") for (int nj = 0; nj < NFRAG; ++nj) {           \
                acc[0][nj] = MF::mfma(a0, bb[nj], acc[0][nj]);               \
                acc[1][nj] = MF::mfma(a1, bb[nj], acc[1][nj]);               \
            }                                                                \
        }                                                                    \
    } while (0)

    // s_waitcnt is wave-uniform: scale-staging waves carry one extra
    // in-flight glds per step
#define B3_BODY(t, b, b2)                                                    \
    do {                                                                     \
        if ((t) + 1 < nchunks) {                                             \
            if (sc_wave) mg_vm_wait<G + 1>();                                \
            else mg_vm_wait<G>();                                            \
        } else mg_vm_wait<0>();                                              \
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                   \
        __builtin_amdgcn_s_barrier();                                        \
        B3_FRAGS(t, b);                                                      \
        if ((t) + 2 < nchunks) B3_STAGE((t) + 2, b2);                        \
    } while (0)

    B3_STAGE(0, 0);
    if (nchunks > 1) B3_STAGE(1, 1);
    int t = 0;
    while (t + 3 <= nchunks) {
        B3_BODY(t + 0, 0, 2);
        B3_BODY(t + 1, 1, 0);
        B3_BODY(t + 2, 2, 1);
        t += 3;
    }
    if (t < nchunks) B3_BODY(t, 0, 2);
    if (t + 1 < nchunks) B3_BODY(t + 1, 1, 0);
#undef B3_BODY
#undef B3_FRAGS
#undef B3_STAGE
    mg_vm_wait<0>();

#pragma unroll
    for (int nj = 0; nj < NFRAG; ++nj) {
        const int n_out = n0 + wn0 + nj * 16 + lrow;
        const float bv =
            bias ? VecTraits<T>::to_float(bias[(long)e * N + n_out]) : 0.f;
#pragma unroll
        for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm0 + qi * 16 + lslot * 4 + r;
                const float v = acc[qi][nj][r] + bv;
                if (combine_out) {
                    const int tk = combine_gather[m];
                    if (tk >= 0) {
                        const float sc =
                            combine_scale ? combine_scale[m] : 1.f;
                        atomicAdd(&combine_out[(long)tk * N + n_out], v * sc);
                    }
                } else {
                    y[(long)m * N + n_out] = VecTraits<T>::from_float(v);
                }
            }
        }
    }
}

// ---------------------------------------------------------------------------
// FP8 (OCP e4m3fn) grouped expert GEMM: weights stored 1 byte/elem with
// ONE fp32 scale per output channel (E, N). Unlike the MXFP4 variant
// (register-staged packed W -> mixed-load vmcnt drains), this keeps the
// ALL-GLDS pipeline — and it uses a DOUBLE-DEPTH K chunk (BK=128): the
// grouped pipeline is chunk-count-bound, not byte-bound (a 64-deep fp8
// chunk moves half the bytes in the SAME time — measured: no speedup),
// so the fp8 kernel stages 128 k-elems per ring step: W is then the
// exact [rows][128B] XOR-swizzled image of the bf16 kernel, x stages as
// TWO 128B half-images, and each step runs 4 MFMA k-slices. Chunk count
// halves; the 2x weight-stream saving becomes wall-clock.
// The B fragment dequantizes on the LDS->register read (4x
// cvt_pk_f32_fp8) and the per-channel scale multiplies the fp32
// accumulator once in the epilogue (scale is constant along K).
// Restrictions: BNT in {128, 256}; K % 128 == 0.


template <typename T>
DEV_INLINE void fp8x8_to_vec8(const unsigned* p2, T* out) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
        const unsigned w = p2[h];
        mg_f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
        mg_f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
        out[4 * h + 0] = VecTraits<T>::from_float(lo[0]);
        out[4 * h + 1] = VecTraits<T>::from_float(lo[1]);
        out[4 * h + 2] = VecTraits<T>::from_float(hi[0]);
        out[4 * h + 3] = VecTraits<T>::from_float(hi[1]);
    }
}

template <typename T, int BNT>
__global__ __launch_bounds__(512, 2) void moe_grouped_fp8_kernel(
    T* __restrict__ y,                   // (Mpad, N)
    const T* __restrict__ x,             // (Mpad, K) — or (T, K) w/ x_gather
    const unsigned char* __restrict__ wq,   // (E, N, K) e4m3fn bytes
    const float* __restrict__ wscale,    // (E, N) per-channel scale
    const int* __restrict__ expert_ids,  // (n_row_tiles,)
    const T* __restrict__ bias,          // (E, N) or null
    const int* __restrict__ x_gather,    // (Mpad,) or null
    float* __restrict__ combine_out,     // (T, N) fp32 or null
    const float* __restrict__ combine_scale,  // (Mpad,)
    const int* __restrict__ combine_gather,   // (Mpad,)
    const int N, const int K) {
    using MF = MgMfma<T>;
    using vec8 = typename MF::vec8;
    constexpr int MT = 64;
    constexpr int BKF = 128;        // k-elems per ring step (double depth)
    constexpr int NWG = BNT / 64;   // W glds per wave per step (8 rows x 128B)
    constexpr int G = 2 + NWG;      // + TWO x half-image glds
    constexpr int NFRAG = BNT / 64;
    constexpr int XBYTES = 2 * MT * 128;   // two 128B half-images
    constexpr int CHUNK_B = XBYTES + BNT * 128;
    static_assert(BNT == 128 || BNT == 256, "fp8 grouped: bn 128/256 only");

    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tile = blockIdx.x;
    const int rt = blockIdx.y;
    const int e = expert_ids[rt];
    if (e < 0) return;
    const int n0 = tile * BNT;
    const int m0 = rt * MT;
    const int nchunks = K / BKF;

    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int wm0 = (wid >> 2) * 32;
    const int wn0 = (wid & 3) * (BNT / 4);
    const int lrow = lane & 15;
    const int lslot = lane >> 4;

    const int xr_lin = m0 + wid * 8 + (lane >> 3);
    const int xr = x_gather ? max(x_gather[xr_lin], 0) : xr_lin;
    const int xslot = (lane & 7) ^ (lane >> 3);
    const char* xsrc_lane = reinterpret_cast<const char*>(x) +
                            (long)xr * K * sizeof(T) + xslot * 16;
    // W rows: 1 byte/elem so a 128-deep chunk is the bf16 kernel's
    // [rows][128B] XOR-swizzled image verbatim
    const char* wsrc[NWG];
#pragma unroll
    for (int i = 0; i < NWG; ++i)
        wsrc[i] = reinterpret_cast<const char*>(wq) +
                  ((long)e * N + n0 + wid * (8 * NWG) + i * 8) * K;

    char* xdst[MG_NBUF];
    char* wdst[MG_NBUF];
#pragma unroll
    for (int b = 0; b < MG_NBUF; ++b) {
        xdst[b] = smem + b * CHUNK_B + wid * 1024;
        wdst[b] = smem + b * CHUNK_B + XBYTES + wid * (NWG * 1024);
    }

    // x half-images: k [0,64) at offset 0, k [64,128) at +MT*128
#define F8_STAGE(t, b)                                                       \
    do {                                                                     \
        const long xoff = (long)(t) * 256;                                   \
        mg_glds_lane128(xsrc_lane + xoff, xdst[b]);                          \
        mg_glds_lane128(xsrc_lane + xoff + 128, xdst[b] + MT * 128);         \
        _Pragma("unroll") for (int i = 0; i < NWG; ++i)                      \
            mg_glds_rows128<2>(wsrc[i] + (long)(t) * 128, (long)K,           \
                               wdst[b] + i * 1024);                          \
    } while (0)

    mg_f32x4 acc[2][NFRAG];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
        for (int nj = 0; nj < NFRAG; ++nj) acc[qi][nj] = {0.f, 0.f, 0.f, 0.f};

    // 4 MFMA k-slices per step: slice kq reads x half-image kq/2 with
    // local ks = kq%2; the W byte group is 16B-slot (kq*2 + lslot/2)
    // XOR-swizzled, byte-half lslot&1
#define F8_FRAGS(b)                                                          \
    do {                                                                     \
        const char* xb = smem + (b) * CHUNK_B;                               \
        const char* wb = xb + XBYTES;                                        \
        _Pragma("unroll") for (int kq = 0; kq < 4; ++kq) {                   \
            const char* xh = xb + (kq >> 1) * (MT * 128);                    \
            const int ks = kq & 1;                                           \
            vec8 a0, a1, bb[NFRAG];                                          \
            {                                                                \
                const int m = wm0 + lrow;                                    \
                const int s = (ks * 4 + lslot) ^ (m & 7);                    \
                a0 = *reinterpret_cast<const vec8*>(xh + m * 128 + s * 16);  \
            }                                                                \
            {                                                                \
                const int m = wm0 + 16 + lrow;                               \
                const int s = (ks * 4 + lslot) ^ (m & 7);                    \
                a1 = *reinterpret_cast<const vec8*>(xh + m * 128 + s * 16);  \
            }                                                                \
            _Pragma("unroll") for (int nj = 0; nj < NFRAG; ++nj) {           \
                const int n = wn0 + nj * 16 + lrow;                          \
                const int s16 = (kq * 2 + (lslot >> 1)) ^ (n & 7);           \
                const unsigned* p2 = reinterpret_cast<const unsigned*>(      \
                    wb + n * 128 + s16 * 16 + (lslot & 1) * 8);              \
                T tmp[8];                                                    \
                fp8x8_to_vec8<T>(p2, tmp);                                   \
                bb[nj] = *reinterpret_cast<const vec8*>(tmp);                \
            }                                                                \
            _Pragma("unroll") for (int nj = 0; nj < NFRAG; ++nj) {           \
                acc[0][nj] = MF::mfma(a0, bb[nj], acc[0][nj]);               \
                acc[1][nj] = MF::mfma(a1, bb[nj], acc[1][nj]);               \
            }                                                                \
        }                                                                    \
    } while (0)

#define F8_BODY(t, b, b2)                                                    \
    do {                                                                     \
        if ((t) + 1 < nchunks) mg_vm_wait<G>();                              \
        else mg_vm_wait<0>();                                                \
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                   \
        __builtin_amdgcn_s_barrier();                                        \
        F8_FRAGS(b);                                                         \
        if ((t) + 2 < nchunks) F8_STAGE((t) + 2, b2);                        \
    } while (0)

    F8_STAGE(0, 0);
    if (nchunks > 1) F8_STAGE(1, 1);
    int t = 0;
    while (t + 3 <= nchunks) {
        F8_BODY(t + 0, 0, 2);
        F8_BODY(t + 1, 1, 0);
        F8_BODY(t + 2, 2, 1);
        t += 3;
    }
    if (t < nchunks) F8_BODY(t, 0, 2);
    if (t + 1 < nchunks) F8_BODY(t + 1, 1, 0);
#undef F8_BODY
#undef F8_FRAGS
#undef F8_STAGE
    mg_vm_wait<0>();

#pragma unroll
    for (int nj = 0; nj < NFRAG; ++nj) {
        const int n_out = n0 + wn0 + nj * 16 + lrow;
        const float wsc = wscale[(long)e * N + n_out];
        const float bv =
            bias ? VecTraits<T>::to_float(bias[(long)e * N + n_out]) : 0.f;
#pragma unroll
        for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm0 + qi * 16 + lslot * 4 + r;
                const float v = acc[qi][nj][r] * wsc + bv;
                if (combine_out) {
                    const int tk = combine_gather[m];
                    if (tk >= 0) {
                        const float sc =
                            combine_scale ? combine_scale[m] : 1.f;
                        atomicAdd(&combine_out[(long)tk * N + n_out], v * sc);
                    }
                } else {
                    y[(long)m * N + n_out] = VecTraits<T>::from_float(v);
                }
            }
        }
    }
}

// ---------------------------------------------------------------------------
// Fused segment-table builder: ONE launch replaces the ~15-kernel torch
// chain (argsort/scatter_add/cumsum/searchsorted — measured 142 us of
// pure launch latency at decode sizes vs ~10 us here). Single workgroup:
// counts via LDS atomics, serial exclusive scans over E (E <= 1024),
// then each assignment claims its within-expert slot with an atomic
// cursor. Order within an expert is therefore NOT the stable argsort
// order — downstream (gather / grouped GEMM / combine) only needs
// order/dest to be a consistent permutation, which any atomic order is
// (fp combine order on GPU is nondeterministic anyway via index_add_).
constexpr int MT_TABLES = 1024;  // threads

__global__ __launch_bounds__(MT_TABLES, 1) void moe_tables_kernel(
    long* __restrict__ order,        // (Ttk,) sorted pos -> assignment idx
    long* __restrict__ dest,         // (Ttk,) sorted pos -> padded row
    int* __restrict__ expert_ids,    // (ntiles,) expert per row tile
    int* __restrict__ gtab,          // (ntiles*tile,) padded row -> token, -1 pad
    const int* __restrict__ idx,     // (Ttk,) expert per assignment
    const int Ttk, const int E, const int ntiles, const int topk,
    const int tile) {
    extern __shared__ int tsm[];
    int* counts = tsm;               // (E,)
    int* starts_raw = tsm + E;       // (E,) exclusive scan of counts
    int* starts_pad = tsm + 2 * E;   // (E,) exclusive scan of padded
    int* cursor = tsm + 3 * E;       // (E,) atomic within-expert rank
    __shared__ int total_pad;

    for (int e = threadIdx.x; e < E; e += blockDim.x) {
        counts[e] = 0;
        cursor[e] = 0;
    }
    __syncthreads();
    for (int i = threadIdx.x; i < Ttk; i += blockDim.x)
        atomicAdd(&counts[idx[i]], 1);
    __syncthreads();
    if (threadIdx.x == 0) {
        int raw = 0, pad = 0;
        for (int e = 0; e < E; ++e) {
            starts_raw[e] = raw;
            starts_pad[e] = pad;
            raw += counts[e];
            pad += (counts[e] + tile - 1) / tile * tile;
        }
        total_pad = pad;
    }
    for (int m = threadIdx.x; m < ntiles * tile; m += blockDim.x)
        gtab[m] = -1;                             // padding rows
    __syncthreads();
    for (int i = threadIdx.x; i < Ttk; i += blockDim.x) {
        const int e = idx[i];
        const int r = atomicAdd(&cursor[e], 1);   // within-expert rank
        const int p = starts_raw[e] + r;          // sorted position
        order[p] = i;
        const int d = starts_pad[e] + r;
        dest[p] = d;
        gtab[d] = i / topk;                       // padded row -> token row
    }
    // expert per row tile: linear search over E (ntiles and E are small)
    for (int t = threadIdx.x; t < ntiles; t += blockDim.x) {
        const int row = t * tile;
        int eid = -1;
        if (row < total_pad) {
            for (int e = 0; e < E; ++e) {
                const int end =
                    starts_pad[e] + (counts[e] + tile - 1) / tile * tile;
                if (row >= starts_pad[e] && row < end && counts[e] > 0) {
                    eid = e;
                    break;
                }
            }
        }
        expert_ids[t] = eid;
    }
}

}  // namespace

void moe_grouped_tables_gpu(torch::Tensor order, torch::Tensor dest,
                            torch::Tensor expert_ids, torch::Tensor gtab,
                            torch::Tensor idx_flat, long num_experts,
                            long topk, long tile) {
    const int Ttk = idx_flat.numel();
    const int E = (int)num_experts;
    const int ntiles = expert_ids.numel();
    TORCH_CHECK(idx_flat.scalar_type() == at::ScalarType::Int &&
                idx_flat.is_contiguous(), "idx_flat int32 contiguous");
    TORCH_CHECK(order.scalar_type() == at::ScalarType::Long &&
                dest.scalar_type() == at::ScalarType::Long &&
                expert_ids.scalar_type() == at::ScalarType::Int &&
                gtab.scalar_type() == at::ScalarType::Int, "out dtypes");
    TORCH_CHECK(order.numel() == Ttk && dest.numel() == Ttk &&
                gtab.numel() == ntiles * tile, "out sizes");
    TORCH_CHECK(tile == 64 || tile == 32, "tile must be 32 or 64");
    TORCH_CHECK(E <= 4096, "num_experts too large for LDS tables");
    const int lds = 4 * E * (int)sizeof(int);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(moe_tables_kernel, dim3(1), dim3(MT_TABLES), lds,
                       stream, order.data_ptr<long>(), dest.data_ptr<long>(),
                       expert_ids.data_ptr<int>(), gtab.data_ptr<int>(),
                       idx_flat.data_ptr<int>(), Ttk, E, ntiles, (int)topk,
                       (int)tile);
    HIP_CHECK_KERNEL();
}

void moe_grouped_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                      torch::Tensor expert_ids,
                      c10::optional<torch::Tensor> bias, long bn,
                      c10::optional<torch::Tensor> x_gather,
                      c10::optional<torch::Tensor> combine_out,
                      c10::optional<torch::Tensor> combine_scale,
                      c10::optional<torch::Tensor> combine_gather, long bk,
                      long mt) {
    TORCH_CHECK(y.dim() == 2 && x.dim() == 2 && w.dim() == 3, "shapes");
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && y.is_contiguous() &&
                expert_ids.is_contiguous(), "contiguous tensors required");
    const int K = x.size(1);
    const int N = w.size(1);
    const int MTT = mt ? (int)mt : 64;
    TORCH_CHECK(MTT == 64 || MTT == 32, "mt must be 32 or 64");
    // with a gather table, x holds the UNSORTED tokens; Mpad comes from
    // the tile table
    const int Mpad = x_gather.has_value() ? (int)expert_ids.numel() * MTT
                                          : (int)x.size(0);
    TORCH_CHECK(w.size(2) == K && y.size(0) == Mpad && y.size(1) == N, "shape");
    TORCH_CHECK(K % MG_BK == 0, "K must be a multiple of 64");
    TORCH_CHECK(Mpad % MTT == 0, "padded rows must be a multiple of mt");
    TORCH_CHECK(expert_ids.scalar_type() == at::ScalarType::Int &&
                expert_ids.numel() == Mpad / MTT,
                "expert_ids must be int (Mpad/mt,)");
    if (x_gather.has_value())
        TORCH_CHECK(x_gather->scalar_type() == at::ScalarType::Int &&
                    x_gather->numel() == Mpad && x_gather->is_contiguous(),
                    "x_gather must be int32 (Mpad,)");
    if (combine_out.has_value()) {
        TORCH_CHECK(combine_gather.has_value(),
                    "combine_out needs combine_gather");
        TORCH_CHECK(combine_out->scalar_type() == at::ScalarType::Float &&
                    combine_out->size(1) == N && combine_out->is_contiguous(),
                    "combine_out must be fp32 (T, N)");
        TORCH_CHECK(combine_gather->scalar_type() == at::ScalarType::Int &&
                    combine_gather->numel() == Mpad, "combine_gather (Mpad,)");
        if (combine_scale.has_value())
            TORCH_CHECK(combine_scale->scalar_type() == at::ScalarType::Float &&
                        combine_scale->numel() == Mpad, "combine_scale (Mpad,)");
    }
    // bn: column-tile width. 0 = auto (widest tile that divides N).
    int BNT = (int)bn;
    if (BNT == 0)
        BNT = (N % 256 == 0) ? 256
              : (N % 192 == 0) ? 192
              : (N % 128 == 0) ? 128 : 64;
    TORCH_CHECK(BNT == 64 || BNT == 128 || BNT == 192 || BNT == 256,
                "bn must be 64/128/192/256");
    TORCH_CHECK(N % BNT == 0, "N must be a multiple of bn");
    // bk: k-depth per ring step. 128 halves the step count (the pipeline
    // is step-count-bound) but its LDS only fits at bn <= 128.
    int BKT = (int)bk;
    if (BKT == 0) BKT = 64;
    TORCH_CHECK(BKT == 64 || BKT == 128, "bk must be 64/128");
    TORCH_CHECK(BKT == 64 || BNT <= 128, "bk=128 needs bn<=128 (LDS)");
    TORCH_CHECK(K % (MG_BK * (BKT / 64)) == 0, "K must be a multiple of bk");

    TORCH_CHECK(MTT == 64 || (BKT == 64 && (BNT == 128 || BNT == 256)),
                "mt=32 needs bk=64 and bn in {128, 256}");
    auto stream = at::hip::getCurrentHIPStream();
    dim3 grid(N / BNT, Mpad / MTT);

#define MG_LAUNCH(T, BNTV, BKTV)                                              \
    do {                                                                      \
        constexpr int LDS =                                                   \
            MG_NBUF * (BKTV / 64) * (64 * 128 + BNTV * 128);                  \
        static bool attr_set = false;                                         \
        if (!attr_set) {                                                      \
            (void)hipFuncSetAttribute(                                        \
                (const void*)moe_grouped_kernel<T, BNTV, BKTV>,               \
                hipFuncAttributeMaxDynamicSharedMemorySize, LDS);             \
            attr_set = true;                                                  \
        }                                                                     \
        hipLaunchKernelGGL((moe_grouped_kernel<T, BNTV, BKTV>), grid,         \
                           dim3(512),                                         \
                           LDS, stream, (T*)y.data_ptr(),                     \
                           (const T*)x.data_ptr(),                            \
                           (const T*)w.data_ptr(), expert_ids.data_ptr<int>(),\
                           bias ? (const T*)bias->data_ptr() : nullptr,       \
                           x_gather ? x_gather->data_ptr<int>() : nullptr,    \
                           combine_out ? combine_out->data_ptr<float>()       \
                                       : nullptr,                             \
                           combine_scale ? combine_scale->data_ptr<float>()   \
                                         : nullptr,                           \
                           combine_gather ? combine_gather->data_ptr<int>()   \
                                          : nullptr,                          \
                           N, K);                                             \
    } while (0)
#define MG_LAUNCH32(T, BNTV)                                                  \
    do {                                                                      \
        constexpr int LDS = MG_NBUF * (32 * 128 + BNTV * 128);                \
        static bool attr_set = false;                                         \
        if (!attr_set) {                                                      \
            (void)hipFuncSetAttribute(                                        \
                (const void*)moe_grouped_kernel<T, BNTV, 64, 32>,             \
                hipFuncAttributeMaxDynamicSharedMemorySize, LDS);             \
            attr_set = true;                                                  \
        }                                                                     \
        hipLaunchKernelGGL((moe_grouped_kernel<T, BNTV, 64, 32>), grid,       \
                           dim3(512),                                         \
                           LDS, stream, (T*)y.data_ptr(),                     \
                           (const T*)x.data_ptr(),                            \
                           (const T*)w.data_ptr(), expert_ids.data_ptr<int>(),\
                           bias ? (const T*)bias->data_ptr() : nullptr,       \
                           x_gather ? x_gather->data_ptr<int>() : nullptr,    \
                           combine_out ? combine_out->data_ptr<float>()       \
                                       : nullptr,                             \
                           combine_scale ? combine_scale->data_ptr<float>()   \
                                         : nullptr,                           \
                           combine_gather ? combine_gather->data_ptr<int>()   \
                                          : nullptr,                          \
                           N, K);                                             \
    } while (0)
#define MG_T(T)                                                               \
    do {                                                                      \
        if (MTT == 32) {                                                      \
            if (BNT == 128) MG_LAUNCH32(T, 128);                              \
            else MG_LAUNCH32(T, 256);                                         \
        } else if (BKT == 128) {                                              \
            if (BNT == 64) MG_LAUNCH(T, 64, 128);                             \
            else MG_LAUNCH(T, 128, 128);                                      \
        } else if (BNT == 64) MG_LAUNCH(T, 64, 64);                           \
        else if (BNT == 128) MG_LAUNCH(T, 128, 64);                           \
        else if (BNT == 192) MG_LAUNCH(T, 192, 64);                           \
        else MG_LAUNCH(T, 256, 64);                                           \
    } while (0)

    if (x.scalar_type() == at::ScalarType::BFloat16) {
        MG_T(__hip_bfloat16);
    } else if (x.scalar_type() == at::ScalarType::Half) {
        MG_T(__half);
    } else {
        TORCH_CHECK(false, "moe_grouped_gemm: bf16/f16 only");
    }
#undef MG_T
#undef MG_LAUNCH32
#undef MG_LAUNCH
    HIP_CHECK_KERNEL();
}

void moe_grouped_gemm_fp8(torch::Tensor y, torch::Tensor x, torch::Tensor wq,
                          torch::Tensor wscale, torch::Tensor expert_ids,
                          c10::optional<torch::Tensor> bias, long bn,
                          c10::optional<torch::Tensor> x_gather,
                          c10::optional<torch::Tensor> combine_out,
                          c10::optional<torch::Tensor> combine_scale,
                          c10::optional<torch::Tensor> combine_gather) {
    TORCH_CHECK(y.dim() == 2 && x.dim() == 2 && wq.dim() == 3 &&
                wscale.dim() == 2, "shapes");
    TORCH_CHECK(x.is_contiguous() && wq.is_contiguous() &&
                wscale.is_contiguous() && y.is_contiguous() &&
                expert_ids.is_contiguous(), "contiguous tensors required");
    const int K = x.size(1);
    const int Mpad = x_gather.has_value() ? (int)expert_ids.numel() * 64
                                          : (int)x.size(0);
    const int N = wq.size(1);
    TORCH_CHECK(wq.size(2) == K && wscale.size(1) == N, "packed shape");
    TORCH_CHECK(y.size(0) == Mpad && y.size(1) == N, "shape");
    TORCH_CHECK(K % 128 == 0 && Mpad % 64 == 0,
                "fp8 grouped: K must be a multiple of 128");
    TORCH_CHECK(wq.scalar_type() == at::ScalarType::Byte &&
                wscale.scalar_type() == at::ScalarType::Float, "wq u8 + f32");
    int BNT = (int)bn;
    if (BNT == 0) BNT = (N % 256 == 0) ? 256 : 128;
    TORCH_CHECK((BNT == 128 || BNT == 256) && N % BNT == 0,
                "fp8 grouped: bn must be 128/256 and divide N");
    if (x_gather.has_value())
        TORCH_CHECK(x_gather->scalar_type() == at::ScalarType::Int &&
                    x_gather->numel() == Mpad, "x_gather (Mpad,) int32");
    if (combine_out.has_value())
        TORCH_CHECK(combine_gather.has_value() &&
                    combine_out->scalar_type() == at::ScalarType::Float &&
                    combine_out->size(1) == N &&
                    combine_gather->numel() == Mpad, "combine args");

    auto stream = at::hip::getCurrentHIPStream();
    dim3 grid(N / BNT, Mpad / 64);

#define MGF_LAUNCH(T, BNTV)                                                   \
    do {                                                                      \
        constexpr int LDS = MG_NBUF * (2 * 64 * 128 + BNTV * 128);                 \
        static bool attr_set = false;                                         \
        if (!attr_set) {                                                      \
            (void)hipFuncSetAttribute(                                        \
                (const void*)moe_grouped_fp8_kernel<T, BNTV>,                 \
                hipFuncAttributeMaxDynamicSharedMemorySize, LDS);             \
            attr_set = true;                                                  \
        }                                                                     \
        hipLaunchKernelGGL((moe_grouped_fp8_kernel<T, BNTV>), grid,           \
                           dim3(512), LDS, stream, (T*)y.data_ptr(),          \
                           (const T*)x.data_ptr(),                            \
                           (const unsigned char*)wq.data_ptr(),               \
                           wscale.data_ptr<float>(),                          \
                           expert_ids.data_ptr<int>(),                        \
                           bias.has_value() ? (const T*)bias->data_ptr() : nullptr, \
                           x_gather ? x_gather->data_ptr<int>() : nullptr,    \
                           combine_out ? combine_out->data_ptr<float>()       \
                                       : nullptr,                             \
                           combine_scale ? combine_scale->data_ptr<float>()   \
                                         : nullptr,                           \
                           combine_gather ? combine_gather->data_ptr<int>()   \
                                          : nullptr,                          \
                           N, K);                                             \
    } while (0)

    if (x.scalar_type() == at::ScalarType::BFloat16) {
        if (BNT == 128) MGF_LAUNCH(__hip_bfloat16, 128);
        else MGF_LAUNCH(__hip_bfloat16, 256);
    } else if (x.scalar_type() == at::ScalarType::Half) {
        if (BNT == 128) MGF_LAUNCH(__half, 128);
        else MGF_LAUNCH(__half, 256);
    } else {
        TORCH_CHECK(false, "moe_grouped_gemm_fp8: bf16/f16 only");
    }
#undef MGF_LAUNCH
    HIP_CHECK_KERNEL();
}

void moe_grouped_gemm_mx4(torch::Tensor y, torch::Tensor x, torch::Tensor wq,
                          torch::Tensor ws, torch::Tensor expert_ids,
                          c10::optional<torch::Tensor> bias, long bn,
                          c10::optional<torch::Tensor> x_gather,
                          c10::optional<torch::Tensor> combine_out,
                          c10::optional<torch::Tensor> combine_scale,
                          c10::optional<torch::Tensor> combine_gather) {
    TORCH_CHECK(y.dim() == 2 && x.dim() == 2 && wq.dim() == 3 && ws.dim() == 3,
                "shapes");
    TORCH_CHECK(x.is_contiguous() && wq.is_contiguous() && ws.is_contiguous() &&
                y.is_contiguous() && expert_ids.is_contiguous(),
                "contiguous tensors required");
    const int K = x.size(1);
    const int Mpad = x_gather.has_value() ? (int)expert_ids.numel() * 64
                                          : (int)x.size(0);
    const int N = wq.size(1);
    if (x_gather.has_value())
        TORCH_CHECK(x_gather->scalar_type() == at::ScalarType::Int &&
                    x_gather->numel() == Mpad, "x_gather (Mpad,) int32");
    if (combine_out.has_value()) {
        TORCH_CHECK(combine_gather.has_value() &&
                    combine_out->scalar_type() == at::ScalarType::Float &&
                    combine_out->size(1) == N &&
                    combine_gather->numel() == Mpad, "combine args");
    }
    TORCH_CHECK(wq.size(2) == K / 2 && ws.size(2) == K / 32, "packed shape");
    TORCH_CHECK(y.size(0) == Mpad && y.size(1) == N, "shape");
    TORCH_CHECK(K % MG_BK == 0 && Mpad % 64 == 0, "align");
    TORCH_CHECK(wq.scalar_type() == at::ScalarType::Byte &&
                ws.scalar_type() == at::ScalarType::Byte, "u8 packed");
    // v3 fast path: all-glds packed pipeline with 256-deep chunks
    // (Mixtral-style dims). Falls through to v1 (register-staged) for
    // K % 256 != 0 (e.g. GPT-OSS K=2880).
    if (K % 256 == 0 && bn == 0 &&
        getenv("NXDI_MX4_V1") == nullptr) {
        const int BNT3 = (N % 128 == 0) ? 128 : 64;
        if (N % BNT3 == 0) {
            auto stream3 = at::hip::getCurrentHIPStream();
            dim3 grid3(N / BNT3, Mpad / 64);
#define MX4B_LAUNCH(T, BNTV)                                                  \
    do {                                                                      \
        constexpr int LDS =                                                   \
            MG_NBUF * (4 * 64 * 128 + BNTV * 128 + BNTV * 16);                \
        static bool attr_set = false;                                         \
        if (!attr_set) {                                                      \
            (void)hipFuncSetAttribute(                                        \
                (const void*)moe_grouped_mx4b_kernel<T, BNTV>,                \
                hipFuncAttributeMaxDynamicSharedMemorySize, LDS);             \
            attr_set = true;                                                  \
        }                                                                     \
        hipLaunchKernelGGL((moe_grouped_mx4b_kernel<T, BNTV>), grid3,         \
                           dim3(512), LDS, stream3, (T*)y.data_ptr(),         \
                           (const T*)x.data_ptr(),                            \
                           (const unsigned char*)wq.data_ptr(),               \
                           (const unsigned char*)ws.data_ptr(),               \
                           expert_ids.data_ptr<int>(),                        \
                           bias.has_value() ? (const T*)bias->data_ptr() : nullptr, \
                           x_gather ? x_gather->data_ptr<int>() : nullptr,    \
                           combine_out ? combine_out->data_ptr<float>()       \
                                       : nullptr,                             \
                           combine_scale ? combine_scale->data_ptr<float>()   \
                                         : nullptr,                           \
                           combine_gather ? combine_gather->data_ptr<int>()   \
                                          : nullptr,                          \
                           N, K);                                             \
    } while (0)
            if (x.scalar_type() == at::ScalarType::BFloat16) {
                if (BNT3 == 128) MX4B_LAUNCH(__hip_bfloat16, 128);
                else MX4B_LAUNCH(__hip_bfloat16, 64);
            } else {
                if (BNT3 == 128) MX4B_LAUNCH(__half, 128);
                else MX4B_LAUNCH(__half, 64);
            }
#undef MX4B_LAUNCH
            HIP_CHECK_KERNEL();
            return;
        }
    }

    int BNT = (int)bn;
    // auto = 64 for mx4: the register-staged packed-W path pays hipcc's
    // mixed-load vmcnt drains once per W row-group, so wide tiles LOSE
    // (measured 531 -> 679 us at gpt-oss T=64 with bn=192); the wide
    // instantiations stay for explicit use
    if (BNT == 0) BNT = 64;
    TORCH_CHECK(BNT == 64 || BNT == 128 || BNT == 192 || BNT == 256,
                "bn must be 64/128/192/256");
    TORCH_CHECK(N % BNT == 0, "N must be a multiple of bn");

    auto stream = at::hip::getCurrentHIPStream();
    dim3 grid(N / BNT, Mpad / 64);

#define MGX_LAUNCH(T, BNTV)                                                   \
    do {                                                                      \
        constexpr int LDS = MG_NBUF * (64 * 128 + BNTV * 128);                \
        static bool attr_set = false;                                         \
        if (!attr_set) {                                                      \
            (void)hipFuncSetAttribute(                                        \
                (const void*)moe_grouped_mx4_kernel<T, BNTV>,                 \
                hipFuncAttributeMaxDynamicSharedMemorySize, LDS);             \
            attr_set = true;                                                  \
        }                                                                     \
        hipLaunchKernelGGL((moe_grouped_mx4_kernel<T, BNTV>), grid,           \
                           dim3(512), LDS, stream, (T*)y.data_ptr(),          \
                           (const T*)x.data_ptr(),                            \
                           (const unsigned char*)wq.data_ptr(),               \
                           (const unsigned char*)ws.data_ptr(),               \
                           expert_ids.data_ptr<int>(),                        \
                           bias.has_value() ? (const T*)bias->data_ptr() : nullptr, \
                           x_gather ? x_gather->data_ptr<int>() : nullptr,    \
                           combine_out ? combine_out->data_ptr<float>()       \
                                       : nullptr,                             \
                           combine_scale ? combine_scale->data_ptr<float>()   \
                                         : nullptr,                           \
                           combine_gather ? combine_gather->data_ptr<int>()   \
                                          : nullptr,                          \
                           N, K);                                             \
    } while (0)
#define MGX_T(T)                                                              \
    do {                                                                      \
        if (BNT == 64) MGX_LAUNCH(T, 64);                                     \
        else if (BNT == 128) MGX_LAUNCH(T, 128);                              \
        else if (BNT == 192) MGX_LAUNCH(T, 192);                              \
        else MGX_LAUNCH(T, 256);                                              \
    } while (0)

    if (x.scalar_type() == at::ScalarType::BFloat16) {
        MGX_T(__hip_bfloat16);
    } else if (x.scalar_type() == at::ScalarType::Half) {
        MGX_T(__half);
    } else {
        TORCH_CHECK(false, "moe_grouped_gemm_mx4: bf16/f16 only");
    }
#undef MGX_T
#undef MGX_LAUNCH
    HIP_CHECK_KERNEL();
}
