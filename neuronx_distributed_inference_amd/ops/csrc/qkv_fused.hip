// Fused post-QKV-GEMM decode plumbing (a big slice of the reference's K2
// fused TKG block, attention_block_tkg): one kernel takes the fused
// qkv GEMM output (B, S, (Hq+2Hkv)*D), applies RoPE to q and k, scatters
// k and v into the linear KV cache at (seq_id, position), and emits q in
// the (B, Hq, S, D) layout the decode-attention kernel wants. Replaces
// 3 transpose-copies + rope + kv_update (5 kernel launches + 4 extra HBM
// round trips per layer per step).
//
// Also: silu/gelu on a fused gate|up GEMM output without the chunk copies.

#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

template <typename T, typename CT>
__global__ void qkv_rope_update_kernel(
    T* __restrict__ q_out,           // (B, Hq, S, D)
    const T* __restrict__ qkv,       // (B, S, (Hq+2Hkv)*D)
    const float* __restrict__ cosb,  // (B, S, D)
    const float* __restrict__ sinb,
    CT* __restrict__ k_cache,        // (Bc, Hkv, Smax, D)
    CT* __restrict__ v_cache,
    const int* __restrict__ positions,  // (B, S)
    const int* __restrict__ seq_ids,    // (B,)
    const int B, const int S, const int Hq, const int Hkv,
    const int D, const int Bc, const int Smax) {
    using VT = VecTraits<T>;
    const int half = D / 2;
    const int Ht = Hq + 2 * Hkv;
    // direct 2D grid: blockIdx.x = (b, s) row, blockIdx.y covers Ht*half —
    // the grid-stride formulation spent its cycles on per-iteration 64-bit
    // div/mod (measured 45 GB/s at prefill scale)
    {
        const int bs = blockIdx.x;
        const int b = bs / S;
        const int s = bs % S;
        const int idx = blockIdx.y * blockDim.x + threadIdx.x;
        if (idx >= Ht * half) return;
        const int h = idx / half;
        const int i = idx % half;

        const T* src = qkv + (((long)b * S + s) * Ht + h) * D;
        const float x1f = VT::to_float(src[i]);
        const float x2f = VT::to_float(src[i + half]);

        float o1f, o2f;
        if (h < Hq + Hkv) {  // rope for q and k
            const float c = cosb[((long)b * S + s) * D + i];
            const float sn = sinb[((long)b * S + s) * D + i];
            o1f = x1f * c - x2f * sn;
            o2f = x2f * c + x1f * sn;
        } else {             // v: plain copy
            o1f = x1f;
            o2f = x2f;
        }

        if (h < Hq) {
            T* dst = q_out + (((long)b * Hq + h) * S + s) * D;
            dst[i] = VT::from_float(o1f);
            dst[i + half] = VT::from_float(o2f);
        } else {
            const int row = seq_ids[b];
            const int pos = positions[(long)b * S + s];
            if (row < 0 || row >= Bc || pos < 0 || pos >= Smax) return;
            const bool is_k = h < Hq + Hkv;
            const int hk = is_k ? h - Hq : h - Hq - Hkv;
            CT* cache = is_k ? k_cache : v_cache;
            CT* dst = cache + (((long)row * Hkv + hk) * Smax + pos) * D;
            dst[i] = VecTraits<CT>::from_float(o1f);
            dst[i + half] = VecTraits<CT>::from_float(o2f);
        }
    }
}

template <typename T, bool GELU>
__global__ void act_mul_fused_kernel(
    T* __restrict__ out,             // (rows, I)
    const T* __restrict__ gate_up,   // (rows, 2I): [gate | up]
    const long rows, const int I) {
    using VT = VecTraits<T>;
    // blockIdx.x = row, blockIdx.y covers I/8 (see qkv kernel note on the
    // div/mod cost of the grid-stride form)
    {
        const long r = blockIdx.x;
        const int c8 = blockIdx.y * blockDim.x + threadIdx.x;
        if (c8 >= I / 8) return;
        Vec8<T> g = load_vec8(gate_up + r * 2 * I + c8 * 8);
        Vec8<T> u = load_vec8(gate_up + r * 2 * I + I + c8 * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float gf = VT::to_float(g.v[j]);
            float uf = VT::to_float(u.v[j]);
            float a;
            if constexpr (GELU) {
                const float cc = 0.7978845608028654f;
                a = 0.5f * gf * (1.f + tanhf(cc * (gf + 0.044715f * gf * gf * gf)));
            } else {
                a = gf / (1.f + __expf(-gf));
            }
            g.v[j] = VT::from_float(a * uf);
        }
        store_vec8(out + r * I + c8 * 8, g);
    }
}

#define DISPATCH_FT(TYPE, NAME, ...)                                           \
    [&] {                                                                      \
        if (TYPE == at::ScalarType::BFloat16) {                                \
            using scalar_t = __hip_bfloat16;                                   \
            return __VA_ARGS__();                                              \
        } else if (TYPE == at::ScalarType::Half) {                             \
            using scalar_t = __half;                                           \
            return __VA_ARGS__();                                              \
        } else if (TYPE == at::ScalarType::Float) {                            \
            using scalar_t = float;                                            \
            return __VA_ARGS__();                                              \
        } else {                                                               \
            TORCH_CHECK(false, NAME ": unsupported dtype");                    \
        }                                                                      \
    }()

void qkv_rope_update(torch::Tensor q_out, torch::Tensor qkv,
                     torch::Tensor cosb, torch::Tensor sinb,
                     torch::Tensor k_cache, torch::Tensor v_cache,
                     torch::Tensor positions, torch::Tensor seq_ids,
                     long Hq, long Hkv) {
    const int B = qkv.size(0), S = qkv.size(1);
    const int D = k_cache.size(3);
    const int Bc = k_cache.size(0), Smax = k_cache.size(2);
    TORCH_CHECK(qkv.size(2) == (Hq + 2 * Hkv) * D, "qkv width mismatch");
    TORCH_CHECK(qkv.is_contiguous(), "qkv must be contiguous");
    const int ht_half = (int)(Hq + 2 * Hkv) * (D / 2);
    dim3 blocks((unsigned)(B * S), (unsigned)((ht_half + 255) / 256));
    auto stream = at::hip::getCurrentHIPStream();
    const bool fp8 = k_cache.scalar_type() == at::ScalarType::Float8_e4m3fn;
    DISPATCH_FT(qkv.scalar_type(), "qkv_rope_update", [&] {
        if (fp8) {
            hipLaunchKernelGGL((qkv_rope_update_kernel<scalar_t, __hip_fp8_e4m3>),
                               blocks, dim3(256), 0, stream,
                               (scalar_t*)q_out.data_ptr(), (const scalar_t*)qkv.data_ptr(),
                               cosb.data_ptr<float>(), sinb.data_ptr<float>(),
                               (__hip_fp8_e4m3*)k_cache.data_ptr(),
                               (__hip_fp8_e4m3*)v_cache.data_ptr(),
                               positions.data_ptr<int>(), seq_ids.data_ptr<int>(),
                               B, S, (int)Hq, (int)Hkv, D, Bc, Smax);
        } else {
            hipLaunchKernelGGL((qkv_rope_update_kernel<scalar_t, scalar_t>),
                               blocks, dim3(256), 0, stream,
                               (scalar_t*)q_out.data_ptr(), (const scalar_t*)qkv.data_ptr(),
                               cosb.data_ptr<float>(), sinb.data_ptr<float>(),
                               (scalar_t*)k_cache.data_ptr(), (scalar_t*)v_cache.data_ptr(),
                               positions.data_ptr<int>(), seq_ids.data_ptr<int>(),
                               B, S, (int)Hq, (int)Hkv, D, Bc, Smax);
        }
    });
    HIP_CHECK_KERNEL();
}

void silu_mul_fused(torch::Tensor out, torch::Tensor gate_up) {
    const int I2 = gate_up.size(-1);
    const long rows = gate_up.numel() / I2;
    const int I = I2 / 2;
    TORCH_CHECK(I % 8 == 0, "intermediate size must be a multiple of 8");
    dim3 blocks((unsigned)rows, (unsigned)((I / 8 + 255) / 256));
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_FT(gate_up.scalar_type(), "silu_mul_fused", [&] {
        hipLaunchKernelGGL((act_mul_fused_kernel<scalar_t, false>), blocks,
                           dim3(256), 0, stream,
                           (scalar_t*)out.data_ptr(), (const scalar_t*)gate_up.data_ptr(),
                           rows, I);
    });
    HIP_CHECK_KERNEL();
}

void gelu_tanh_mul_fused(torch::Tensor out, torch::Tensor gate_up) {
    const int I2 = gate_up.size(-1);
    const long rows = gate_up.numel() / I2;
    const int I = I2 / 2;
    TORCH_CHECK(I % 8 == 0, "intermediate size must be a multiple of 8");
    dim3 blocks((unsigned)rows, (unsigned)((I / 8 + 255) / 256));
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_FT(gate_up.scalar_type(), "gelu_tanh_mul_fused", [&] {
        hipLaunchKernelGGL((act_mul_fused_kernel<scalar_t, true>), blocks,
                           dim3(256), 0, stream,
                           (scalar_t*)out.data_ptr(), (const scalar_t*)gate_up.data_ptr(),
                           rows, I);
    });
    HIP_CHECK_KERNEL();
}
