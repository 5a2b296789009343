// Memory-bound elementwise/normalization kernels for gfx950.
//
// RMSNorm (K7-analog), fused residual-add RMSNorm, SiLU*up / GELU*up
// (the K5 MLP activation seam) and RoPE apply (rotate-half convention).
// All bf16/f16 paths use 16-byte vectorized loads (guide G13: scalar bf16
// loads cost ~2x); RMSNorm is two-pass with the second pass served from L1/L2.

#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

// ---------------------------------------------------------------------------
// RMSNorm: one 256-thread workgroup per row, grid-stride over rows.
// ---------------------------------------------------------------------------
template <typename T, bool FUSED_ADD>
__global__ void rmsnorm_kernel(
    T* __restrict__ out, const T* __restrict__ x, T* __restrict__ residual,
    const T* __restrict__ weight, const float eps, const int rows, const int H) {
    __shared__ float red[16];
    using VT = VecTraits<T>;
    for (int row = blockIdx.x; row < rows; row += gridDim.x) {
        const T* xr = x + (long)row * H;
        T* rr = FUSED_ADD ? residual + (long)row * H : nullptr;
        float ssq = 0.f;
        for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
            Vec8<T> v = load_vec8(xr + i);
            if constexpr (FUSED_ADD) {
                Vec8<T> r = load_vec8(rr + i);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    float f = VT::to_float(v.v[j]) + VT::to_float(r.v[j]);
                    v.v[j] = VT::from_float(f);
                    ssq += f * f;
                }
                store_vec8(rr + i, v);  // residual stream updated in place
            } else {
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    float f = VT::to_float(v.v[j]);
                    ssq += f * f;
                }
            }
        }
        ssq = block_reduce_sum(ssq, red);
        const float inv = rsqrtf(ssq / H + eps);
        const T* src = FUSED_ADD ? rr : xr;
        T* orow = out + (long)row * H;
        for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
            Vec8<T> v = load_vec8(src + i);
            Vec8<T> w = load_vec8(weight + i);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                v.v[j] = VT::from_float(VT::to_float(v.v[j]) * inv * VT::to_float(w.v[j]));
            store_vec8(orow + i, v);
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// SiLU(gate) * up and GELU-tanh(gate) * up
// ---------------------------------------------------------------------------
template <typename T, bool GELU>
__global__ void act_mul_kernel(
    T* __restrict__ out, const T* __restrict__ gate, const T* __restrict__ up,
    const long n) {
    using VT = VecTraits<T>;
    const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    const long stride = (long)gridDim.x * blockDim.x * 8;
    for (long i = i0; i < n; i += stride) {
        Vec8<T> g = load_vec8(gate + i);
        Vec8<T> u = load_vec8(up + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float gf = VT::to_float(g.v[j]);
            float uf = VT::to_float(u.v[j]);
            float a;
            if constexpr (GELU) {
                // tanh approximation of GELU
                const float c = 0.7978845608028654f;  // sqrt(2/pi)
                a = 0.5f * gf * (1.f + tanhf(c * (gf + 0.044715f * gf * gf * gf)));
            } else {
                a = gf / (1.f + __expf(-gf));
            }
            g.v[j] = VT::from_float(a * uf);
        }
        store_vec8(out + i, g);
    }
}

// ---------------------------------------------------------------------------
// RoPE apply (rotate-half): q (B,Hq,S,D), k (B,Hkv,S,D) in place.
// cos/sin: (B,S,D) fp32. Each thread handles the pair (i, i+D/2), i < D/2;
// lanes cover consecutive i so both halves load coalesced.
// ---------------------------------------------------------------------------
template <typename T>
__global__ void rope_kernel(
    T* __restrict__ q, T* __restrict__ k,
    const float* __restrict__ cosb, const float* __restrict__ sinb,
    const int B, const int Hq, const int Hkv, const int S, const int D) {
    using VT = VecTraits<T>;
    const int half = D / 2;
    const long total = (long)B * (Hq + Hkv) * S * half;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total; idx += stride) {
        const int i = idx % half;
        long rest = idx / half;
        const int s = rest % S;
        rest /= S;
        const int h = rest % (Hq + Hkv);
        const int b = rest / (Hq + Hkv);
        T* base = (h < Hq)
            ? q + (((long)b * Hq + h) * S + s) * D
            : k + (((long)b * Hkv + (h - Hq)) * S + s) * D;
        const float c = cosb[((long)b * S + s) * D + i];
        const float sn = sinb[((long)b * S + s) * D + i];
        const float x1 = VT::to_float(base[i]);
        const float x2 = VT::to_float(base[i + half]);
        base[i] = VT::from_float(x1 * c - x2 * sn);
        base[i + half] = VT::from_float(x2 * c + x1 * sn);
    }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------
#define DISPATCH_FLOAT_TYPES(TYPE, NAME, ...)                                  \
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

static int ew_grid(long work_items, int block) {
    long blocks = (work_items + block - 1) / block;
    if (blocks > 2048) blocks = 2048;  // grid-stride the rest (guide G11)
    if (blocks < 1) blocks = 1;
    return (int)blocks;
}

void rms_norm(torch::Tensor out, torch::Tensor x, torch::Tensor weight, double eps) {
    const int H = x.size(-1);
    const long rows = x.numel() / H;
    TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_FLOAT_TYPES(x.scalar_type(), "rms_norm", [&] {
        hipLaunchKernelGGL(
            (rmsnorm_kernel<scalar_t, false>), dim3(ew_grid(rows, 1)), dim3(256), 0, stream,
            (scalar_t*)out.data_ptr(), (const scalar_t*)x.data_ptr(), nullptr,
            (const scalar_t*)weight.data_ptr(), (float)eps, (int)rows, H);
    });
    HIP_CHECK_KERNEL();
}

void rms_norm_fused_add(torch::Tensor out, torch::Tensor x, torch::Tensor residual,
                        torch::Tensor weight, double eps) {
    const int H = x.size(-1);
    const long rows = x.numel() / H;
    TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_FLOAT_TYPES(x.scalar_type(), "rms_norm_fused_add", [&] {
        hipLaunchKernelGGL(
            (rmsnorm_kernel<scalar_t, true>), dim3(ew_grid(rows, 1)), dim3(256), 0, stream,
            (scalar_t*)out.data_ptr(), (const scalar_t*)x.data_ptr(),
            (scalar_t*)residual.data_ptr(), (const scalar_t*)weight.data_ptr(),
            (float)eps, (int)rows, H);
    });
    HIP_CHECK_KERNEL();
}

void silu_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up) {
    const long n = gate.numel();
    TORCH_CHECK(n % 8 == 0, "numel must be a multiple of 8");
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_FLOAT_TYPES(gate.scalar_type(), "silu_mul", [&] {
        hipLaunchKernelGGL((act_mul_kernel<scalar_t, false>),
                           dim3(ew_grid(n / 8, 256)), dim3(256), 0, stream,
                           (scalar_t*)out.data_ptr(), (const scalar_t*)gate.data_ptr(),
                           (const scalar_t*)up.data_ptr(), n);
    });
    HIP_CHECK_KERNEL();
}

void gelu_tanh_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up) {
    const long n = gate.numel();
    TORCH_CHECK(n % 8 == 0, "numel must be a multiple of 8");
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_FLOAT_TYPES(gate.scalar_type(), "gelu_tanh_mul", [&] {
        hipLaunchKernelGGL((act_mul_kernel<scalar_t, true>),
                           dim3(ew_grid(n / 8, 256)), dim3(256), 0, stream,
                           (scalar_t*)out.data_ptr(), (const scalar_t*)gate.data_ptr(),
                           (const scalar_t*)up.data_ptr(), n);
    });
    HIP_CHECK_KERNEL();
}

void rope_apply(torch::Tensor q, torch::Tensor k, torch::Tensor cosb, torch::Tensor sinb) {
    TORCH_CHECK(q.dim() == 4 && k.dim() == 4, "q/k must be (B,H,S,D)");
    const int B = q.size(0), Hq = q.size(1), S = q.size(2), D = q.size(3);
    const int Hkv = k.size(1);
    TORCH_CHECK(cosb.scalar_type() == at::ScalarType::Float, "cos/sin must be fp32");
    TORCH_CHECK(cosb.dim() == 3 && cosb.size(2) == D, "cos must be (B,S,D)");
    const long total = (long)B * (Hq + Hkv) * S * (D / 2);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_FLOAT_TYPES(q.scalar_type(), "rope_apply", [&] {
        hipLaunchKernelGGL((rope_kernel<scalar_t>),
                           dim3(ew_grid(total, 256)), dim3(256), 0, stream,
                           (scalar_t*)q.data_ptr(), (scalar_t*)k.data_ptr(),
                           cosb.data_ptr<float>(), sinb.data_ptr<float>(),
                           B, Hq, Hkv, S, D);
    });
    HIP_CHECK_KERNEL();
}
