// Causal flash-attention prefill for gfx950 (K1/K11-analog; ref
// nkilib attention_cte called at attention_base.py:722-744 and the in-repo
// sliding-window NKI kernel modules/sliding_window/attention.py:38-477).
//
// MFMA (16x16x32 bf16/f16) flash forward with online softmax:
//   * workgroup = 4 waves = one 128-row Q block of one (b, hq); each wave
//     owns 32 q rows. K/V tiles (KVBLK=32 keys) staged in LDS by the whole
//     workgroup; V is transposed during staging so the PV B-fragments read
//     16 contiguous bytes per lane.
//   * all LDS tiles use padded row strides (row_bytes = data + 16) so the
//     B-fragment column reads are bank-conflict-free (guide §6 G4: row-major
//     [KV][128] bf16 is an up-to-16-way conflict).
//   * P round-trips through LDS between QK^T (C layout) and PV (A layout).
//   * causal + sliding-window masking on the fp32 scores; GQA native (the
//     KV head is staged once for the gq query heads that share it —
//     workgroups differ in hq but hit the same K/V through L2).
//
// This is the correctness-first structure (guide §5 ladder step ~2-3);
// planned upgrades: K-tile XOR swizzle (T2), async-stage split (T14),
// double buffering, exp2 folding.

#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

template <typename T> struct MfmaTraits;
template <> struct MfmaTraits<__hip_bfloat16> {
    using vec8 = bf16x8;
    static DEV_INLINE f32x4 mfma(vec8 a, vec8 b, f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    }
};
template <> struct MfmaTraits<__half> {
    using vec8 = f16x8;
    static DEV_INLINE f32x4 mfma(vec8 a, vec8 b, f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
    }
};

constexpr int QBLK = 128;       // q rows per workgroup
constexpr int WQ = 32;          // q rows per wave
constexpr int KVBLK = 64;       // keys per LDS tile
constexpr float NEG_INF = -1e30f;

template <typename T, int D>
__global__ __launch_bounds__(256, 2) void flash_prefill_kernel(
    T* __restrict__ out,       // (B, Hq, S, D)
    const T* __restrict__ q,   // (B, Hq, S, D)
    const T* __restrict__ k,   // (B, Hkv, S, D)
    const T* __restrict__ v,   // (B, Hkv, S, D)
    const float scale, const int window,
    const int Hq, const int Hkv, const int S,
    const float* __restrict__ sinks,   // (Hq,) learned sink logits or null
    const int* __restrict__ seq_ids,   // non-null: K/V live in the KV cache
    const int Smax) {                  // cache row capacity (= S when direct)
    using MT = MfmaTraits<T>;
    using vec8 = typename MT::vec8;
    constexpr int DI = D / 16;       // O col fragments per wave
    constexpr int KS = D / 32;       // K-dim steps for QK^T
    // padded LDS row strides (bytes)
    constexpr int K_STRIDE = D * 2 + 16;
    constexpr int VT_STRIDE = KVBLK * 2 + 16;
    constexpr int P_STRIDE = KVBLK * 2 + 16;
    // XOR swizzle on the V^T image in-row offset, keyed by the V^T row (the
    // head-dim d). The transpose staging writes one b16 per (d, key) from 16
    // threads whose d differ by 8 at byte stride VT_STRIDE (a multiple of
    // 16B) — all on one bank without this (measured: SQ_LDS_BANK_CONFLICT
    // 2.5x SQ_BUSY_CYCLES). The swizzle spreads them over 8 bank slots; the
    // b128 PV reads stay 16B-aligned and at worst 2-way.
#define VT_SWZ(row) ((((row) >> 3) & 7) << 4)

    __shared__ __attribute__((aligned(16))) char smem[
        KVBLK * K_STRIDE + D * VT_STRIDE + 4 * WQ * P_STRIDE];
    char* k_lds = smem;
    char* vt_lds = smem + KVBLK * K_STRIDE;
    char* p_lds_all = vt_lds + D * VT_STRIDE;

    const int qblk = blockIdx.x;
    const int hq = blockIdx.y;
    const int b = blockIdx.z;
    const int hkv = hq / (Hq / Hkv);
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    char* p_lds = p_lds_all + wid * WQ * P_STRIDE;

    const int q_lo = qblk * QBLK;
    const int wq_lo = q_lo + wid * WQ;          // this wave's first q row
    const int wq_hi = min(wq_lo + WQ, S) - 1;   // last valid q row (abs)

    const long q_base = (((long)b * Hq + hq) * S) * D;
    // seq_ids[b] can be -1 (ModelWrapper pads CTE batches with seq_id -1);
    // clamp to row 0 so the cached path never reads before the allocation —
    // pad-row outputs are discarded by the host anyway.
    const long kv_row = (seq_ids != nullptr) ? max((long)seq_ids[b], 0L) : b;
    const long kv_base = ((kv_row * Hkv + hkv) * (long)Smax) * D;

    // ---- Q fragments into registers (A layout: row = lane&15, k-elems
    // (lane>>4)*8 .. +8, per (qi, ks)) ----
    vec8 q_frag[2][KS];
    const int qa_row = lane & 15;
    const int qa_off = (lane >> 4) * 8;
#pragma unroll
    for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
        for (int ks = 0; ks < KS; ++ks) {
            const int row = min(wq_lo + qi * 16 + qa_row, S - 1);
            q_frag[qi][ks] = *reinterpret_cast<const vec8*>(
                q + q_base + (long)row * D + ks * 32 + qa_off);
        }
    }

    f32x4 o_acc[2][DI];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
        for (int di = 0; di < DI; ++di) o_acc[qi][di] = {0.f, 0.f, 0.f, 0.f};
    // per-lane softmax state: 4 rows per C fragment, 2 qi fragments
    float m_run[2][4], l_run[2][4];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            m_run[qi][r] = NEG_INF;
            l_run[qi][r] = 0.f;
        }

    const int kv_hi = min(S, q_lo + QBLK);  // causal: no keys beyond block end
    int kv_start = 0;
    if (window > 0) kv_start = max(0, (q_lo - window + 1) / KVBLK * KVBLK);
    // window < 0: chunked attention (chunk C = -window); the block's first
    // query bounds the earliest reachable row at its chunk start
    else if (window < 0) kv_start = (q_lo / -window) * -window / KVBLK * KVBLK;

    // Staging note: occupancy (not staging latency) is the binding
    // constraint — keep live state minimal so 2 waves/SIMD fit (the
    // register-prefetch variant measured null because it pushed the kernel
    // to 329 regs = 1 wave/SIMD).
    constexpr int NIT = (KVBLK * D) / (256 * 8);
    const int st_row = (threadIdx.x * 8) / D;        // this thread's key row
    const int st_d = (threadIdx.x * 8) % D;          // and d offset
    const int row_step = (256 * 8) / D;

    for (int k0 = kv_start; k0 < kv_hi; k0 += KVBLK) {
#pragma unroll
        for (int it = 0; it < NIT; ++it) {
            const int row = st_row + it * row_step;
            const int krow = min(k0 + row, S - 1);
            Vec8<T> kv8 = load_vec8(k + kv_base + (long)krow * D + st_d);
            Vec8<T> vv8 = load_vec8(v + kv_base + (long)krow * D + st_d);
            *reinterpret_cast<int4*>(k_lds + row * K_STRIDE + st_d * 2) =
                *reinterpret_cast<int4*>(kv8.v);
#pragma unroll
            for (int j = 0; j < 8; ++j)  // transpose: V^T[d+j][row]
                *reinterpret_cast<T*>(
                    vt_lds + (st_d + j) * VT_STRIDE + ((row * 2) ^ VT_SWZ(st_d + j))) =
                    vv8.v[j];
        }
        __syncthreads();

        if (k0 <= wq_hi) {  // this wave still has un-masked keys in the tile
            // ---- QK^T: S_tile[32q][KVBLK k] as 2 x KVBLK/16 fragments ----
            constexpr int KI = KVBLK / 16;
            f32x4 s_frag[2][KI];
#pragma unroll
            for (int qi = 0; qi < 2; ++qi)
#pragma unroll
                for (int ki = 0; ki < KI; ++ki) {
                    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
                    for (int ks = 0; ks < KS; ++ks) {
                        // B fragment: col = key (lane&15), elems = dims
                        const vec8 kb = *reinterpret_cast<const vec8*>(
                            k_lds + (ki * 16 + (lane & 15)) * K_STRIDE +
                            (ks * 32 + (lane >> 4) * 8) * 2);
                        acc = MT::mfma(q_frag[qi][ks], kb, acc);
                    }
                    s_frag[qi][ki] = acc;
                }

            // ---- mask + online softmax (C layout: col=lane&15 is the key,
            // row=(lane>>4)*4+reg is the q) ----
            // Tiles fully below the causal diagonal (and inside the window /
            // sequence bounds) skip every per-element mask op — that is most
            // tiles, and the masking VALU was a large share of wave cycles.
            const bool full_tile =
                (k0 + KVBLK - 1 <= wq_lo) && (k0 + KVBLK <= S)
                && (window <= 0 || k0 >= wq_lo + WQ - window)
                && (window >= 0 || k0 >= ((wq_lo + WQ - 1) / -window) * -window);
            float p_new[2][4];  // per-qi per-reg row max of this tile
#pragma unroll
            for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int q_abs = wq_lo + qi * 16 + (lane >> 4) * 4 + r;
                    float mx = NEG_INF;
                    if (full_tile) {
#pragma unroll
                        for (int ki = 0; ki < KI; ++ki) {
                            const float sv = s_frag[qi][ki][r] * scale;
                            s_frag[qi][ki][r] = sv;
                            mx = fmaxf(mx, sv);
                        }
                    } else
#pragma unroll
                    for (int ki = 0; ki < KI; ++ki) {
                        const int k_abs = k0 + ki * 16 + (lane & 15);
                        float sv = s_frag[qi][ki][r] * scale;
                        const bool ok = (k_abs <= q_abs) && (k_abs < S) && (q_abs < S)
                            && (window <= 0 || k_abs > q_abs - window)
                            && (window >= 0 || k_abs >= (q_abs / -window) * -window);
                        sv = ok ? sv : NEG_INF;
                        s_frag[qi][ki][r] = sv;
                        mx = fmaxf(mx, sv);
                    }
                    // row max across the 16 key-columns
#pragma unroll
                    for (int off = 8; off > 0; off >>= 1)
                        mx = fmaxf(mx, __shfl_xor(mx, off, WAVE_SIZE));
                    p_new[qi][r] = mx;
                }
            }
#pragma unroll
            for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const float m_new = fmaxf(m_run[qi][r], p_new[qi][r]);
                    const float corr = __expf(m_run[qi][r] - m_new);
                    float row_sum = 0.f;
                    if (full_tile) {
#pragma unroll
                        for (int ki = 0; ki < KI; ++ki) {
                            const float p = __expf(s_frag[qi][ki][r] - m_new);
                            s_frag[qi][ki][r] = p;
                            row_sum += p;
                        }
                    } else
#pragma unroll
                    for (int ki = 0; ki < KI; ++ki) {
                        const float sv = s_frag[qi][ki][r];
                        const float p = (sv <= NEG_INF) ? 0.f : __expf(sv - m_new);
                        s_frag[qi][ki][r] = p;
                        row_sum += p;
                    }
#pragma unroll
                    for (int off = 8; off > 0; off >>= 1)
                        row_sum += __shfl_xor(row_sum, off, WAVE_SIZE);
                    l_run[qi][r] = l_run[qi][r] * corr + row_sum;
                    m_run[qi][r] = m_new;
                    // rescale O rows
#pragma unroll
                    for (int di = 0; di < DI; ++di)
                        o_acc[qi][di][r] *= corr;
                }
            }

            // ---- P -> LDS (bf16, A-layout-readable padded rows) ----
#pragma unroll
            for (int qi = 0; qi < 2; ++qi)
#pragma unroll
                for (int ki = 0; ki < KI; ++ki)
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        const int qrow = qi * 16 + (lane >> 4) * 4 + r;
                        const int kcol = ki * 16 + (lane & 15);
                        *reinterpret_cast<T*>(p_lds + qrow * P_STRIDE + kcol * 2) =
                            VecTraits<T>::from_float(s_frag[qi][ki][r]);
                    }
            // lgkmcnt(0): this wave's P writes land before its own reads.
            // P is per-wave LDS so no workgroup barrier is needed; the
            // "memory" clobber keeps the compiler from hoisting the reads.
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

            // ---- PV: O[32q][D] += P[32q][KVBLK] * V[KVBLK][D], in 32-key
            // chunks (one A fragment per chunk) ----
#pragma unroll
            for (int kc2 = 0; kc2 < KVBLK / 32; ++kc2) {
#pragma unroll
                for (int qi = 0; qi < 2; ++qi) {
                    const vec8 pa = *reinterpret_cast<const vec8*>(
                        p_lds + (qi * 16 + (lane & 15)) * P_STRIDE +
                        (kc2 * 32 + (lane >> 4) * 8) * 2);
#pragma unroll
                    for (int di = 0; di < DI; ++di) {
                        const int vrow = di * 16 + (lane & 15);
                        const vec8 vb = *reinterpret_cast<const vec8*>(
                            vt_lds + vrow * VT_STRIDE +
                            (((kc2 * 32 + (lane >> 4) * 8) * 2) ^ VT_SWZ(vrow)));
                        o_acc[qi][di] = MT::mfma(pa, vb, o_acc[qi][di]);
                    }
                }
            }
        }
        __syncthreads();  // compute done before the next write_tile overwrites
    }

    // ---- epilogue: normalize and store ----
#pragma unroll
    for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int q_abs = wq_lo + qi * 16 + (lane >> 4) * 4 + r;
            if (q_abs >= S) continue;
            float l_tot = l_run[qi][r];
            if (sinks != nullptr)  // sink joins the denominator only
                l_tot += __expf(sinks[hq] - m_run[qi][r]);
            const float inv = (l_tot > 0.f) ? 1.f / l_tot : 0.f;
#pragma unroll
            for (int di = 0; di < DI; ++di) {
                const int dcol = di * 16 + (lane & 15);
                out[q_base + (long)q_abs * D + dcol] =
                    VecTraits<T>::from_float(o_acc[qi][di][r] * inv);
            }
        }
    }
}

// NOTE on the PV B-fragment: mfma_f32_16x16x32 B is a 32(K)x16(N) operand —
// lane holds col n = lane&15 and k-elems (lane>>4)*8..+7. Our read above
// loads V^T[n_dim][k] rows, giving lane {d = di*16 + lane&15, k = (lane>>4)*8+j}
// which matches B[k][n] with n=d. Same shape logic for the QK^T B (K tile).

static void flash_prefill_impl(torch::Tensor& out, torch::Tensor& q,
                               const torch::Tensor& k, const torch::Tensor& v,
                               double scale, long window,
                               c10::optional<torch::Tensor>& sinks_opt,
                               const int* seq_ids_ptr, int Smax_arg, int Hkv) {
    const int B = q.size(0), Hq = q.size(1), S = q.size(2), D = q.size(3);
    TORCH_CHECK(Hq % Hkv == 0, "Hq must divide by Hkv");
    TORCH_CHECK(D == 64 || D == 128, "flash_prefill: head_dim 64/128 only");
    dim3 grid((S + QBLK - 1) / QBLK, Hq, B);
    auto stream = at::hip::getCurrentHIPStream();
    torch::Tensor sinks_t;
    const float* sinks = nullptr;
    if (sinks_opt.has_value()) {
        sinks_t = sinks_opt->to(torch::kFloat32).contiguous();
        sinks = sinks_t.data_ptr<float>();
    }
    if (q.scalar_type() == at::ScalarType::BFloat16) {
        if (D == 128)
            hipLaunchKernelGGL((flash_prefill_kernel<__hip_bfloat16, 128>), grid, dim3(256), 0,
                               stream, (__hip_bfloat16*)out.data_ptr(),
                               (const __hip_bfloat16*)q.data_ptr(),
                               (const __hip_bfloat16*)k.data_ptr(),
                               (const __hip_bfloat16*)v.data_ptr(),
                               (float)scale, (int)window, Hq, Hkv, S, sinks,
                               seq_ids_ptr, Smax_arg);
        else
            hipLaunchKernelGGL((flash_prefill_kernel<__hip_bfloat16, 64>), grid, dim3(256), 0,
                               stream, (__hip_bfloat16*)out.data_ptr(),
                               (const __hip_bfloat16*)q.data_ptr(),
                               (const __hip_bfloat16*)k.data_ptr(),
                               (const __hip_bfloat16*)v.data_ptr(),
                               (float)scale, (int)window, Hq, Hkv, S, sinks,
                               seq_ids_ptr, Smax_arg);
    } else if (q.scalar_type() == at::ScalarType::Half) {
        if (D == 128)
            hipLaunchKernelGGL((flash_prefill_kernel<__half, 128>), grid, dim3(256), 0,
                               stream, (__half*)out.data_ptr(), (const __half*)q.data_ptr(),
                               (const __half*)k.data_ptr(), (const __half*)v.data_ptr(),
                               (float)scale, (int)window, Hq, Hkv, S, sinks,
                               seq_ids_ptr, Smax_arg);
        else
            hipLaunchKernelGGL((flash_prefill_kernel<__half, 64>), grid, dim3(256), 0,
                               stream, (__half*)out.data_ptr(), (const __half*)q.data_ptr(),
                               (const __half*)k.data_ptr(), (const __half*)v.data_ptr(),
                               (float)scale, (int)window, Hq, Hkv, S, sinks,
                               seq_ids_ptr, Smax_arg);
    } else {
        TORCH_CHECK(false, "flash_prefill: bf16/f16 only");
    }
    HIP_CHECK_KERNEL();
}

void flash_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                   torch::Tensor v, double scale, long window,
                   c10::optional<torch::Tensor> sinks_opt) {
    flash_prefill_impl(out, q, k, v, scale, window, sinks_opt, nullptr,
                       (int)q.size(2), (int)k.size(1));
}

void flash_prefill_cached(torch::Tensor out, torch::Tensor q,
                          torch::Tensor k_cache, torch::Tensor v_cache,
                          torch::Tensor seq_ids, double scale, long window,
                          c10::optional<torch::Tensor> sinks_opt) {
    // K/V read from the (Bc, Hkv, Smax, D) cache rows just written —
    // prefill straight off the cache, no fresh K/V tensors
    auto sid = seq_ids.to(torch::kInt32).contiguous();
    flash_prefill_impl(out, q, k_cache, v_cache, scale, window, sinks_opt,
                       sid.data_ptr<int>(), (int)k_cache.size(2),
                       (int)k_cache.size(1));
}
