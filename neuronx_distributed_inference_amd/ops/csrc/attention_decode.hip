// Token-generation (decode) attention for gfx950 — the K2-analog
// (ref attention_block_tkg, modules/attention/attention_base.py:1186-1381).
//
// Memory-bound split-KV design for the linear cache (B,Hkv,Smax,D):
//  * pass 1: grid (nsplit, Hkv*Sq, B), 256-thread workgroups (4 waves).
//    Lane groups of D/8 lanes stream K/V rows with 16-B loads (8 bf16 per
//    lane = one ds-free register path; guide G13). Each group computes the
//    q.k dots for all GQ query heads of its KV head at once, so the KV
//    bytes are read ONCE per (b, hkv) regardless of GQA ratio, online
//    softmax in registers, cross-group combine via __shfl_xor butterflies,
//    cross-wave combine via LDS, per-split partials (m, l, o) to workspace.
//  * pass 2: combines the nsplit partials per (b, hq, sq) row.
//
// The split over the sequence gives the chip >> 256 workgroups at decode
// batch sizes (256 CUs in 8 XCDs need large grids) and is the same
// structure flash-decoding (seq-sharded KV, ref C5) extends across GPUs.

#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

constexpr float M_INIT = -1e30f;

// PAGED=false: linear cache (Bc, Hkv, Smax, D), row l at seq row seq_ids[b].
// PAGED=true: paged cache (num_blocks, block_size, Hkv, D) walked through
// block_table (B, max_blocks); Smax is the per-sequence capacity
// max_blocks*block_size.
// CT = cache storage type (T itself, or fp8-e4m3 when the KV cache is
// quantized; the k-scale is folded into the softmax scale by the host and
// the v-scale applied in pass 2).
// 2-way packed dot (v_dot2): halves the QK VALU work and the q-register
// footprint — the GQ=8 (70B TP1) case was occupancy-bound at 178 VGPRs.
typedef __bf16 dot2_bf16x2 __attribute__((ext_vector_type(2)));
typedef _Float16 dot2_f16x2 __attribute__((ext_vector_type(2)));
template <typename T> struct Dot2 {
    static constexpr bool ok = false;
    using v2 = float;
    static __device__ __forceinline__ float dot(v2, v2, float c) { return c; }
    static __device__ __forceinline__ v2 pack(float, float) { return 0.f; }
};
template <> struct Dot2<__hip_bfloat16> {
    static constexpr bool ok = true;
    using v2 = dot2_bf16x2;
    static __device__ __forceinline__ float dot(v2 a, v2 b, float c) {
        return __builtin_amdgcn_fdot2_f32_bf16(a, b, c, false);
    }
    static __device__ __forceinline__ v2 pack(float a, float b) {
        v2 r; r[0] = (__bf16)a; r[1] = (__bf16)b; return r;
    }
};
template <> struct Dot2<__half> {
    static constexpr bool ok = true;
    using v2 = dot2_f16x2;
    static __device__ __forceinline__ float dot(v2 a, v2 b, float c) {
        return __builtin_amdgcn_fdot2(a, b, c, false);
    }
    static __device__ __forceinline__ v2 pack(float a, float b) {
        v2 r; r[0] = (_Float16)a; r[1] = (_Float16)b; return r;
    }
};

template <typename T, typename CT, int D, int GQ, bool PAGED>
__global__ __launch_bounds__(256) void decode_attn_pass1(
    float* __restrict__ o_part,        // (B, Hq, Sq, nsplit, D)
    float* __restrict__ ml_part,       // (B, Hq, Sq, nsplit, 2)
    const T* __restrict__ q,           // (B, Hq, Sq, D) — Hq = Hkv*gq
    const CT* __restrict__ k_cache,    // (Bc, Hkv, Smax, D)
    const CT* __restrict__ v_cache,
    const int* __restrict__ positions, // (B, Sq) per-query absolute position
    const int* __restrict__ seq_ids,   // (B,) (linear) / unused (paged)
    const int* __restrict__ block_table,  // (B, max_blocks) when PAGED
    const float scale, const int window, const int ring,
    const int Hkv, const int Sq, const int gq,
    const int Smax, const int chunk, const int nsplit,
    const int block_size, const int max_blocks) {
    using VT = VecTraits<T>;
    constexpr int GS = D / 8;          // lanes per group
    constexpr int NG = 256 / GS;       // groups per workgroup

    const int split = blockIdx.x;
    const int hkv = blockIdx.y / Sq;
    const int sq = blockIdx.y % Sq;
    const int b = blockIdx.z;
    const int Hq = Hkv * gq;

    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wid = threadIdx.x / WAVE_SIZE;
    const int grp = threadIdx.x / GS;         // 0..NG-1
    const int c = threadIdx.x % GS;           // lane-in-group

    const int row = seq_ids[b];
    // query at absolute position p attends cache rows [0, p]
    const int eff_len = positions[(long)b * Sq + sq] + 1;
    int lo = 0;
    int span_hi = eff_len;
    int pmax = 0;
    if (ring) {
        // RING cache (compact sliding-window layers): slot s holds the
        // newest position == s (mod Smax); all written slots are scanned
        // and masked by true position below. pmax = max over VALID (>=0)
        // positions of the row — ragged chunked-prefill batches carry
        // trailing -1 pads (CPU ref ops/__init__.py uses positions.amax).
        pmax = -1;
        for (int s = 0; s < Sq; ++s)
            pmax = max(pmax, positions[(long)b * Sq + s]);
        span_hi = min(pmax + 1, Smax);
    } else if (window > 0 && eff_len > window) {
        lo = eff_len - window;
    }
    // window < 0: Llama4 CHUNKED attention with chunk size -window — the
    // query attends only rows in its own chunk [(p/C)*C, p]
    else if (window < 0) lo = ((eff_len - 1) / (-window)) * (-window);
    const int split_lo = max(lo, split * chunk);
    const int split_hi = min(span_hi, (split + 1) * chunk);

    // ---- q into registers, pre-scaled ----
    constexpr bool PK = Dot2<T>::ok && std::is_same<T, CT>::value;
    using D2 = Dot2<T>;
    typename D2::v2 q_pk[PK ? GQ : 1][4];
    float q_reg[PK ? 1 : GQ][8];
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
        if (g < gq) {
            const long qoff = ((((long)b * Hq) + hkv * gq + g) * Sq + sq) * D + c * 8;
            Vec8<T> qv = load_vec8(q + qoff);
            if constexpr (PK) {
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    q_pk[g][j] = D2::pack(VT::to_float(qv.v[2 * j]) * scale,
                                          VT::to_float(qv.v[2 * j + 1]) * scale);
            } else {
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    q_reg[g][j] = VT::to_float(qv.v[j]) * scale;
            }
        }
    }

    float m[GQ], lsum[GQ], o_acc[GQ][8];
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
        m[g] = M_INIT;
        lsum[g] = 0.f;
#pragma unroll
        for (int j = 0; j < 8; ++j) o_acc[g][j] = 0.f;
    }

    // ---- stream K/V rows (next row's loads issued before computing the
    // current one so the HBM latency hides under the dot/softmax VALU) ----
    if ((PAGED || row >= 0) && split_lo < split_hi) {
        const CT* kbase = PAGED ? k_cache : k_cache + (((long)row * Hkv + hkv) * Smax) * D;
        const CT* vbase = PAGED ? v_cache : v_cache + (((long)row * Hkv + hkv) * Smax) * D;
        // unconditional prefetch with clamped addresses — a conditional
        // ("load or keep") makes hipcc branch around the loads and wait
        // vmcnt(0) per iteration (guide §5 trap 4(c))
        auto row_addr = [&](int l, bool& valid) -> long {
            if constexpr (PAGED) {
                const int blk = block_table[(long)b * max_blocks + l / block_size];
                valid = blk >= 0;
                return (((long)max(blk, 0) * block_size + l % block_size) * Hkv + hkv) * D + c * 8;
            }
            valid = true;
            return (long)l * D + c * 8;
        };
        // per-group CONTIGUOUS row runs: a group streams rows
        // [g_lo, g_hi) back to back, so each group is one sequential HBM
        // stream (the strided l += NG layout read 256B chunks at 4 KB
        // stride — DRAM-page/channel hostile, same pathology measured on
        // the skinny GEMM). Online softmax is order-free, so the group->row
        // assignment is irrelevant to the math.
        const int span = split_hi - split_lo;
        const int rpg = (span + NG - 1) / NG;
        const int g_lo = split_lo + grp * rpg;
        const int g_hi = min(g_lo + rpg, split_hi);
        // row-compute body (shared by the 2-deep software pipeline legs)
        auto attn_row = [&](int l, bool valid_cur, const Vec8<CT>& kv_cur,
                            const Vec8<CT>& vv_cur) {
            if (PAGED && !valid_cur) return;
            if (ring) {
                const int p_s = pmax - ((pmax - l) % Smax);  // slot's true pos
                if (p_s >= eff_len ||
                    (window > 0 && p_s <= eff_len - 1 - window))
                    return;
            }
            const Vec8<CT>& vv_use = vv_cur;

            float s[GQ];
            if constexpr (PK) {
                const typename D2::v2* kp =
                    reinterpret_cast<const typename D2::v2*>(kv_cur.v);
#pragma unroll
                for (int g = 0; g < GQ; ++g) {
                    if (g >= gq) continue;
                    float d = 0.f;
#pragma unroll
                    for (int j = 0; j < 4; ++j) d = D2::dot(q_pk[g][j], kp[j], d);
                    s[g] = group_reduce_sum<GS>(d);
                }
            } else {
                float kf[8];
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    kf[j] = VecTraits<CT>::to_float(kv_cur.v[j]);
#pragma unroll
                for (int g = 0; g < GQ; ++g) {
                    if (g >= gq) continue;
                    float d = 0.f;
#pragma unroll
                    for (int j = 0; j < 8; ++j) d += q_reg[g][j] * kf[j];
                    s[g] = group_reduce_sum<GS>(d);  // all GS lanes get the dot
                }
            }
#pragma unroll
            for (int g = 0; g < GQ; ++g) {
                if (g >= gq) continue;
                const float mn = fmaxf(m[g], s[g]);
                const float corr = __expf(m[g] - mn);
                const float p = __expf(s[g] - mn);
                lsum[g] = lsum[g] * corr + p;
                m[g] = mn;
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    o_acc[g][j] = o_acc[g][j] * corr + p * VecTraits<CT>::to_float(vv_use.v[j]);
            }
        };

        // 2-deep pipeline: two row buffers alternate, so at each wait
        // point the OTHER buffer's load is still in flight and the
        // waited-on load has had ~2 iterations to land (depth-1 exposed
        // most of the HBM latency per row: the register copy forces the
        // vmcnt wait before the next load can issue).
        // Clamps: empty ranges and tails re-load a safe row.
        bool v0, v1;
        long off0 = row_addr(min(g_lo, split_hi - 1), v0);
        Vec8<CT> k0 = load_vec8(kbase + off0);
        Vec8<CT> w0 = load_vec8(vbase + off0);
        long off1 = row_addr((g_lo + 1 < g_hi) ? g_lo + 1
                                               : min(g_lo, split_hi - 1), v1);
        Vec8<CT> k1 = load_vec8(kbase + off1);
        Vec8<CT> w1 = load_vec8(vbase + off1);
        for (int l = g_lo; l < g_hi; l += 2) {
            {
                const bool vc = v0;
                const Vec8<CT> kc = k0, wc = w0;
                const int ln = (l + 2 < g_hi) ? l + 2 : l;
                off0 = row_addr(ln, v0);
                k0 = load_vec8(kbase + off0);
                w0 = load_vec8(vbase + off0);
                attn_row(l, vc, kc, wc);
            }
            if (l + 1 < g_hi) {
                const bool vc = v1;
                const Vec8<CT> kc = k1, wc = w1;
                const int ln = (l + 3 < g_hi) ? l + 3 : l + 1;
                off1 = row_addr(ln, v1);
                k1 = load_vec8(kbase + off1);
                w1 = load_vec8(vbase + off1);
                attn_row(l + 1, vc, kc, wc);
            }
        }
    }

    // ---- combine groups within the wave (xor butterfly over group bits) ----
#pragma unroll
    for (int off = GS; off < WAVE_SIZE; off <<= 1) {
#pragma unroll
        for (int g = 0; g < GQ; ++g) {
            if (g >= gq) continue;
            const float m_o = __shfl_xor(m[g], off, WAVE_SIZE);
            const float l_o = __shfl_xor(lsum[g], off, WAVE_SIZE);
            const float mn = fmaxf(m[g], m_o);
            const float f_s = __expf(m[g] - mn);
            const float f_o = __expf(m_o - mn);
            lsum[g] = lsum[g] * f_s + l_o * f_o;
            m[g] = mn;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float oo = __shfl_xor(o_acc[g][j], off, WAVE_SIZE);
                o_acc[g][j] = o_acc[g][j] * f_s + oo * f_o;
            }
        }
    }

    // ---- combine the 4 waves through LDS ----
    __shared__ float o_lds[4][GQ > 4 ? GQ : 4][D];
    __shared__ float ml_lds[4][GQ > 4 ? GQ : 4][2];
    if (lane < GS) {
#pragma unroll
        for (int g = 0; g < GQ; ++g) {
            if (g >= gq) continue;
#pragma unroll
            for (int j = 0; j < 8; ++j) o_lds[wid][g][c * 8 + j] = o_acc[g][j];
            if (c == 0) {
                ml_lds[wid][g][0] = m[g];
                ml_lds[wid][g][1] = lsum[g];
            }
        }
    }
    __syncthreads();
    if (wid == 0 && lane < GS) {
        for (int g = 0; g < gq; ++g) {
            float mm = M_INIT, ll = 0.f, oo[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) oo[j] = 0.f;
            for (int w = 0; w < 4; ++w) {
                const float mw = ml_lds[w][g][0];
                const float lw = ml_lds[w][g][1];
                const float mn = fmaxf(mm, mw);
                const float f_s = __expf(mm - mn);
                const float f_w = __expf(mw - mn);
                ll = ll * f_s + lw * f_w;
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    oo[j] = oo[j] * f_s + o_lds[w][g][c * 8 + j] * f_w;
                mm = mn;
            }
            const long base =
                ((((long)b * Hq + hkv * gq + g) * Sq + sq) * nsplit + split);
#pragma unroll
            for (int j = 0; j < 8; ++j) o_part[base * D + c * 8 + j] = oo[j];
            if (c == 0) {
                ml_part[base * 2 + 0] = mm;
                ml_part[base * 2 + 1] = ll;
            }
        }
    }
}

template <typename T, int D>
__global__ void decode_attn_pass2(
    T* __restrict__ out,               // (B, Hq, Sq, D)
    const float* __restrict__ o_part,  // (B, Hq, Sq, nsplit, D)
    const float* __restrict__ ml_part, // (B, Hq, Sq, nsplit, 2)
    const int nsplit, const float v_scale,
    const float* __restrict__ sinks,   // (Hq,) learned sink logits or null
    const int Hq, const int Sq,
    float* __restrict__ lse_out) {     // (B,Hq,Sq) log-sum-exp, or null —
                                       // the cross-rank flash-decoding handle
    using VT = VecTraits<T>;
    const long rowid = blockIdx.x;  // over B*Hq*Sq
    const int lane = threadIdx.x;
    // global max over splits
    float mstar = M_INIT;
    for (int i = 0; i < nsplit; ++i)
        mstar = fmaxf(mstar, ml_part[(rowid * nsplit + i) * 2 + 0]);
    float L = 0.f;
    if (sinks != nullptr) {  // sink joins the softmax denominator only
        const int h = (int)((rowid / Sq) % Hq);
        mstar = fmaxf(mstar, sinks[h]);
        L += __expf(sinks[h] - mstar);
    }
    for (int i = 0; i < nsplit; ++i) {
        const float mi = ml_part[(rowid * nsplit + i) * 2 + 0];
        const float li = ml_part[(rowid * nsplit + i) * 2 + 1];
        L += li * __expf(mi - mstar);
    }
    const float inv = (L > 0.f) ? v_scale / L : 0.f;
    if (lse_out != nullptr && lane == 0)
        lse_out[rowid] = (L > 0.f) ? mstar + __logf(L) : -INFINITY;
#pragma unroll
    for (int d = lane; d < D; d += WAVE_SIZE) {
        float acc = 0.f;
        for (int i = 0; i < nsplit; ++i) {
            const float mi = ml_part[(rowid * nsplit + i) * 2 + 0];
            acc += o_part[(rowid * nsplit + i) * D + d] * __expf(mi - mstar);
        }
        out[rowid * D + d] = VT::from_float(acc * inv);
    }
}

// ---------------------------------------------------------------------------
template <typename T, typename CT, int D, int GQ>
static void launch_decode(torch::Tensor& out, const torch::Tensor& q,
                          const torch::Tensor& k_cache, const torch::Tensor& v_cache,
                          const torch::Tensor& seq_lens, const torch::Tensor& seq_ids,
                          float scale, int window, int gq,
                          const torch::Tensor* block_table = nullptr,
                          int block_size = 0, float v_scale = 1.0f,
                          const torch::Tensor* sinks = nullptr,
                          torch::Tensor* lse_out = nullptr,
                          int ring = 0) {
    const int B = q.size(0), Hq = q.size(1), Sq = q.size(2);
    const bool paged = block_table != nullptr;
    const int Hkv = paged ? k_cache.size(2) : k_cache.size(1);
    const int max_blocks = paged ? (int)block_table->size(1) : 0;
    const int Smax = paged ? max_blocks * block_size : (int)k_cache.size(2);
    // split count: fill the chip (>=1024 workgroups when possible), chunks of
    // >=128 rows so per-split work amortizes.
    int nsplit = 1024 / std::max(1, B * Hkv * Sq);
    nsplit = std::max(1, std::min({nsplit, (Smax + 127) / 128, 64}));
    const int chunk = (Smax + nsplit - 1) / nsplit;

    auto opts = torch::TensorOptions().dtype(torch::kFloat32).device(q.device());
    auto o_part = torch::empty({(long)B * Hq * Sq * nsplit * D}, opts);
    auto ml_part = torch::empty({(long)B * Hq * Sq * nsplit * 2}, opts);
    auto stream = at::hip::getCurrentHIPStream();

    dim3 grid1(nsplit, Hkv * Sq, B);
    if (paged) {
        hipLaunchKernelGGL((decode_attn_pass1<T, CT, D, GQ, true>), grid1, dim3(256), 0, stream,
                           o_part.data_ptr<float>(), ml_part.data_ptr<float>(),
                           (const T*)q.data_ptr(), (const CT*)k_cache.data_ptr(),
                           (const CT*)v_cache.data_ptr(), seq_lens.data_ptr<int>(),
                           seq_ids.data_ptr<int>(), block_table->data_ptr<int>(),
                           scale, window, 0, Hkv, Sq, gq, Smax, chunk, nsplit,
                           block_size, max_blocks);
    } else {
        hipLaunchKernelGGL((decode_attn_pass1<T, CT, D, GQ, false>), grid1, dim3(256), 0, stream,
                           o_part.data_ptr<float>(), ml_part.data_ptr<float>(),
                           (const T*)q.data_ptr(), (const CT*)k_cache.data_ptr(),
                           (const CT*)v_cache.data_ptr(), seq_lens.data_ptr<int>(),
                           seq_ids.data_ptr<int>(), nullptr,
                           scale, window, ring, Hkv, Sq, gq, Smax, chunk, nsplit, 0, 0);
    }
    HIP_CHECK_KERNEL();
    dim3 grid2(B * Hq * Sq);
    hipLaunchKernelGGL((decode_attn_pass2<T, D>), grid2, dim3(WAVE_SIZE), 0, stream,
                       (T*)out.data_ptr(), o_part.data_ptr<float>(),
                       ml_part.data_ptr<float>(), nsplit, v_scale,
                       sinks ? sinks->data_ptr<float>() : nullptr, Hq, Sq,
                       lse_out ? lse_out->data_ptr<float>() : nullptr);
    HIP_CHECK_KERNEL();
}

template <typename T, int D>
static void dispatch_gq(torch::Tensor& out, const torch::Tensor& q,
                        const torch::Tensor& k_cache, const torch::Tensor& v_cache,
                        const torch::Tensor& seq_lens, const torch::Tensor& seq_ids,
                        float scale, int window, int gq,
                        const torch::Tensor* bt = nullptr, int bs = 0,
                        float v_scale = 1.0f, const torch::Tensor* sinks = nullptr,
                        torch::Tensor* lse_out = nullptr, int ring = 0) {
    const bool fp8 = k_cache.scalar_type() == at::ScalarType::Float8_e4m3fn;
#define GQ_CASE(CT)                                                                     \
    do {                                                                                \
        if (gq <= 1)                                                                    \
            launch_decode<T, CT, D, 1>(out, q, k_cache, v_cache, seq_lens, seq_ids,     \
                                       scale, window, gq, bt, bs, v_scale, sinks,       \
                                       lse_out, ring);                                  \
        else if (gq <= 2)                                                               \
            launch_decode<T, CT, D, 2>(out, q, k_cache, v_cache, seq_lens, seq_ids,     \
                                       scale, window, gq, bt, bs, v_scale, sinks,       \
                                       lse_out, ring);                                  \
        else if (gq <= 4)                                                               \
            launch_decode<T, CT, D, 4>(out, q, k_cache, v_cache, seq_lens, seq_ids,     \
                                       scale, window, gq, bt, bs, v_scale, sinks,       \
                                       lse_out, ring);                                  \
        else if (gq <= 8)                                                               \
            launch_decode<T, CT, D, 8>(out, q, k_cache, v_cache, seq_lens, seq_ids,     \
                                       scale, window, gq, bt, bs, v_scale, sinks,       \
                                       lse_out, ring);                                  \
        else                                                                            \
            TORCH_CHECK(false, "GQA ratio > 8 not supported yet (got ", gq, ")");       \
    } while (0)
    if (fp8) {
        GQ_CASE(__hip_fp8_e4m3);
    } else {
        GQ_CASE(T);
    }
#undef GQ_CASE
}

void attention_decode_paged(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor seq_lens, torch::Tensor block_table,
                            double scale, long window) {
    // paged cache (num_blocks, block_size, Hkv, D); `seq_lens` carries
    // per-query POSITIONS, shape (B, Sq)
    TORCH_CHECK(q.dim() == 4 && k_cache.dim() == 4, "bad shapes");
    TORCH_CHECK(seq_lens.numel() == q.size(0) * q.size(2),
                "positions must be (B, Sq)");
    const int D = q.size(3);
    const int Hq = q.size(1), Hkv = k_cache.size(2);
    const int bs = k_cache.size(1);
    TORCH_CHECK(Hq % Hkv == 0, "Hq must be a multiple of Hkv");
    const int gq = Hq / Hkv;
    auto seq_ids = seq_lens;  // unused in the paged path

#define CASE_P(T)                                                               \
    do {                                                                        \
        if (D == 128)                                                           \
            dispatch_gq<T, 128>(out, q, k_cache, v_cache, seq_lens, seq_ids,    \
                                (float)scale, (int)window, gq, &block_table, bs); \
        else if (D == 64)                                                       \
            dispatch_gq<T, 64>(out, q, k_cache, v_cache, seq_lens, seq_ids,     \
                               (float)scale, (int)window, gq, &block_table, bs); \
        else                                                                    \
            TORCH_CHECK(false, "paged decode: head_dim 64/128 only");           \
    } while (0)

    if (q.scalar_type() == at::ScalarType::BFloat16) {
        CASE_P(__hip_bfloat16);
    } else if (q.scalar_type() == at::ScalarType::Half) {
        CASE_P(__half);
    } else if (q.scalar_type() == at::ScalarType::Float) {
        CASE_P(float);
    } else {
        TORCH_CHECK(false, "attention_decode_paged: unsupported dtype");
    }
#undef CASE_P
}

void attention_decode(torch::Tensor out, torch::Tensor q,
                      torch::Tensor k_cache, torch::Tensor v_cache,
                      torch::Tensor seq_lens, torch::Tensor seq_ids,
                      double scale, long window,
                      c10::optional<torch::Tensor> sinks_opt,
                      c10::optional<torch::Tensor> lse_opt,
                      long ring) {
    // lse_opt: fp32 (B, Hq, Sq), filled with per-query log-sum-exp — the
    // cross-rank flash-decoding combine handle
    // `seq_lens` carries per-query POSITIONS, shape (B, Sq)
    TORCH_CHECK(q.dim() == 4 && k_cache.dim() == 4, "bad shapes");
    TORCH_CHECK(seq_lens.numel() == q.size(0) * q.size(2),
                "positions must be (B, Sq)");
    const int D = q.size(3);
    const int Hq = q.size(1), Hkv = k_cache.size(1);
    TORCH_CHECK(Hq % Hkv == 0, "Hq must be a multiple of Hkv");
    const int gq = Hq / Hkv;
    TORCH_CHECK(q.is_contiguous() && k_cache.is_contiguous() && v_cache.is_contiguous(),
                "contiguous tensors required");
    torch::Tensor sinks_t;
    const torch::Tensor* sinks = nullptr;
    if (sinks_opt.has_value()) {
        sinks_t = sinks_opt->to(torch::kFloat32).contiguous();
        sinks = &sinks_t;
    }
    torch::Tensor* lse_ptr = nullptr;
    torch::Tensor lse_t;
    if (lse_opt.has_value()) {
        lse_t = *lse_opt;
        TORCH_CHECK(lse_t.scalar_type() == torch::kFloat32 &&
                    lse_t.numel() == (long)q.size(0) * Hq * q.size(2),
                    "lse must be fp32 (B,Hq,Sq)");
        lse_ptr = &lse_t;
    }

#define CASE_T(T)                                                               \
    do {                                                                        \
        if (D == 128)                                                           \
            dispatch_gq<T, 128>(out, q, k_cache, v_cache, seq_lens, seq_ids,    \
                                (float)scale, (int)window, gq, nullptr, 0,      \
                                1.0f, sinks, lse_ptr, (int)ring);               \
        else if (D == 64)                                                       \
            dispatch_gq<T, 64>(out, q, k_cache, v_cache, seq_lens, seq_ids,     \
                               (float)scale, (int)window, gq, nullptr, 0,       \
                               1.0f, sinks, lse_ptr, (int)ring);                \
        else if (D == 32)                                                       \
            dispatch_gq<T, 32>(out, q, k_cache, v_cache, seq_lens, seq_ids,     \
                               (float)scale, (int)window, gq, nullptr, 0,       \
                               1.0f, sinks, lse_ptr, (int)ring);                \
        else if (D == 256)                                                      \
            dispatch_gq<T, 256>(out, q, k_cache, v_cache, seq_lens, seq_ids,    \
                                (float)scale, (int)window, gq, nullptr, 0,      \
                                1.0f, sinks, lse_ptr, (int)ring);               \
        else                                                                    \
            TORCH_CHECK(false, "head_dim must be 32/64/128/256, got ", D);      \
    } while (0)

    if (q.scalar_type() == at::ScalarType::BFloat16) {
        CASE_T(__hip_bfloat16);
    } else if (q.scalar_type() == at::ScalarType::Half) {
        CASE_T(__half);
    } else if (q.scalar_type() == at::ScalarType::Float) {
        CASE_T(float);
    } else {
        TORCH_CHECK(false, "attention_decode: unsupported dtype");
    }
#undef CASE_T
}
