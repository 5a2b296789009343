// FP8 (OCP e4m3) activation quantization + scaled-GEMM epilogue.
//
// K6-analog (ref rmsnorm_quant_kernel, modeling_llama.py:553-575): the
// torch-op formulation of dynamic per-tensor activation quant materialized
// x in fp32 twice and issued ~6 launches per linear, which made the fp8
// weight path NET SLOWER than bf16 (profiles/r01_bench_summary.md). Here:
//
// - amax pass: vec8 bf16 reads, wave-reduced, one fp32 atomicMax (positive
//   floats are monotonic as uint32 bit patterns);
// - cast pass: reads the amax scalar on device (scale = amax/448), writes
//   fp8 and the scale tensor torch._scaled_mm consumes — no host sync, so
//   the whole chain is hipGraph-capturable;
// - epilogue: y32 (M,N) * w_row_scale (N) [+ bias] -> bf16 in one kernel
//   (the per-row weight scale cannot ride through scaled_mm's per-tensor
//   scale_b on this stack).

#include "common.h"

#define DISPATCH_FT(TYPE, NAME, ...)                                           \
    [&] {                                                                      \
        switch (TYPE) {                                                        \
            case at::ScalarType::BFloat16: {                                   \
                using scalar_t = __hip_bfloat16;                               \
                return __VA_ARGS__();                                          \
            }                                                                  \
            case at::ScalarType::Half: {                                       \
                using scalar_t = __half;                                       \
                return __VA_ARGS__();                                          \
            }                                                                  \
            case at::ScalarType::Float: {                                      \
                using scalar_t = float;                                        \
                return __VA_ARGS__();                                          \
            }                                                                  \
            default:                                                           \
                TORCH_CHECK(false, NAME ": unsupported dtype");                \
        }                                                                      \
    }()

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr float kFp8Max = 448.0f;
constexpr int kAmaxBlocks = 128;

// stage 1: per-block partial maxima — NO atomics (2048 waves hammering one
// atomic address serialized at ~39 us/call; profiles/r01 fp8 analysis)
template <typename T>
__global__ void amax_kernel(const T* __restrict__ x, const long n8,
                            float* __restrict__ partial) {
    using VT = VecTraits<T>;
    float m = 0.f;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
         i += stride) {
        Vec8<T> v = load_vec8(x + i * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) m = fmaxf(m, fabsf(VT::to_float(v.v[j])));
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_xor(m, off, 64));
    __shared__ float red[4];
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = m;
    __syncthreads();
    if (threadIdx.x == 0)
        partial[blockIdx.x] = fmaxf(fmaxf(red[0], red[1]),
                                    fmaxf(red[2], red[3]));
}

// stage 2 fused into the cast: every block folds the 128 partials itself
template <typename T>
__global__ void fp8_cast_kernel(const T* __restrict__ x,
                                __hip_fp8_e4m3* __restrict__ out,
                                const float* __restrict__ partial,
                                float* __restrict__ scale_out, const long n8) {
    using VT = VecTraits<T>;
    float m = (threadIdx.x < kAmaxBlocks) ? partial[threadIdx.x] : 0.f;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_xor(m, off, 64));
    __shared__ float red[4];
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = m;
    __syncthreads();
    const float amax = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
    const float scale = fmaxf(amax, 1e-8f) / kFp8Max;
    const float inv = 1.0f / scale;
    if (blockIdx.x == 0 && threadIdx.x == 0) *scale_out = scale;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
         i += stride) {
        Vec8<T> v = load_vec8(x + i * 8);
        Vec8<__hip_fp8_e4m3> q;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = VT::to_float(v.v[j]) * inv;
            f = fminf(fmaxf(f, -kFp8Max), kFp8Max);
            q.v[j] = VecTraits<__hip_fp8_e4m3>::from_float(f);
        }
        store_vec8(out + i * 8, q);
    }
}

template <typename T>
__global__ void fp8_epilogue_kernel(T* __restrict__ out,
                                    const float* __restrict__ y32,
                                    const float* __restrict__ w_scale,  // (N,)
                                    const T* __restrict__ bias,         // or null
                                    const long rows, const int N) {
    const long r = blockIdx.x;
    for (int c = blockIdx.y * blockDim.x + threadIdx.x; c < N;
         c += gridDim.y * blockDim.x) {
        float v = y32[r * N + c] * w_scale[c];
        if (bias != nullptr) v += VecTraits<T>::to_float(bias[c]);
        out[r * N + c] = VecTraits<T>::from_float(v);
    }
}

}  // namespace

void fp8_quant(torch::Tensor x, torch::Tensor out, torch::Tensor scale,
               torch::Tensor amax_ws) {
    TORCH_CHECK(x.is_contiguous() && x.numel() % 8 == 0);
    TORCH_CHECK(out.scalar_type() == at::ScalarType::Float8_e4m3fn);
    TORCH_CHECK(amax_ws.scalar_type() == torch::kFloat32);
    const long n8 = x.numel() / 8;
    auto stream = at::hip::getCurrentHIPStream();
    TORCH_CHECK(amax_ws.numel() >= kAmaxBlocks, "amax workspace too small");
    const long cast_blocks = std::min((n8 + 255) / 256, (long)2048);
    DISPATCH_FT(x.scalar_type(), "fp8_quant", [&] {
        hipLaunchKernelGGL((amax_kernel<scalar_t>), dim3(kAmaxBlocks), dim3(256),
                           0, stream, (const scalar_t*)x.data_ptr(), n8,
                           amax_ws.data_ptr<float>());
        hipLaunchKernelGGL((fp8_cast_kernel<scalar_t>), dim3((int)cast_blocks),
                           dim3(256), 0, stream, (const scalar_t*)x.data_ptr(),
                           (__hip_fp8_e4m3*)out.data_ptr(),
                           amax_ws.data_ptr<float>(), scale.data_ptr<float>(),
                           n8);
    });
    HIP_CHECK_KERNEL();
}

void fp8_epilogue(torch::Tensor out, torch::Tensor y32, torch::Tensor w_scale,
                  c10::optional<torch::Tensor> bias_opt) {
    const int N = out.size(-1);
    const long rows = out.numel() / N;
    TORCH_CHECK(y32.scalar_type() == torch::kFloat32);
    auto stream = at::hip::getCurrentHIPStream();
    dim3 grid((unsigned)rows, (unsigned)std::min((N + 255) / 256, 16));
    torch::Tensor bias_t;
    const void* bias = nullptr;
    if (bias_opt.has_value()) {
        bias_t = bias_opt->contiguous();
        bias = bias_t.data_ptr();
    }
    DISPATCH_FT(out.scalar_type(), "fp8_epilogue", [&] {
        hipLaunchKernelGGL((fp8_epilogue_kernel<scalar_t>), grid, dim3(256), 0,
                           stream, (scalar_t*)out.data_ptr(),
                           y32.data_ptr<float>(), w_scale.data_ptr<float>(),
                           (const scalar_t*)bias, rows, N);
    });
    HIP_CHECK_KERNEL();
}
