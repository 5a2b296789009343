// KV-cache scatter-write kernels (K13/K16-analog).
//
// Linear cache:  k_cache/v_cache (Bc, Hkv, Smax, D); new K/V (B, Hkv, Sn, D)
// written at per-(b,s) positions with a seq_ids row map (continuous
// batching; out-of-range row => skipped garbage write, ref
// data_parallel_kv_cache_manager.py).
// Paged cache:   (num_blocks, block_size, Hkv, D) written by flat slot ids.

#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

template <typename T, typename CT>
__global__ void kv_update_kernel(
    CT* __restrict__ k_cache, CT* __restrict__ v_cache,
    const T* __restrict__ k_new, const T* __restrict__ v_new,
    const int* __restrict__ positions,   // (B, Sn)
    const int* __restrict__ seq_ids,     // (B,)
    const int B, const int Hkv, const int Sn, const int D,
    const int Bc, const int Smax) {
    // one vec8 element per thread over (B, Hkv, Sn, D/8)
    const int dv = D / 8;
    const long total = (long)B * Hkv * Sn * dv;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total; idx += stride) {
        const int d8 = idx % dv;
        long rest = idx / dv;
        const int s = rest % Sn;
        rest /= Sn;
        const int h = rest % Hkv;
        const int b = rest / Hkv;
        const int row = seq_ids[b];
        if (row < 0 || row >= Bc) continue;
        const int pos = positions[b * Sn + s];
        if (pos < 0 || pos >= Smax) continue;
        const long src = (((long)b * Hkv + h) * Sn + s) * D + d8 * 8;
        const long dst = (((long)row * Hkv + h) * Smax + pos) * D + d8 * 8;
        store_vec8(k_cache + dst, convert_vec8<CT, T>(load_vec8(k_new + src)));
        store_vec8(v_cache + dst, convert_vec8<CT, T>(load_vec8(v_new + src)));
    }
}

template <typename T, typename CT>
__global__ void block_kv_update_kernel(
    CT* __restrict__ k_cache, CT* __restrict__ v_cache,
    const T* __restrict__ k_new, const T* __restrict__ v_new,
    const int* __restrict__ slot_mapping,  // (N,)
    const int N, const int Hkv, const int D, const long nslots) {
    const int dv = D / 8;
    const long total = (long)N * Hkv * dv;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total; idx += stride) {
        const int d8 = idx % dv;
        long rest = idx / dv;
        const int h = rest % Hkv;
        const int n = rest / Hkv;
        const int slot = slot_mapping[n];
        if (slot < 0 || slot >= nslots) continue;
        const long src = ((long)n * Hkv + h) * D + d8 * 8;
        const long dst = ((long)slot * Hkv + h) * D + d8 * 8;
        store_vec8(k_cache + dst, convert_vec8<CT, T>(load_vec8(k_new + src)));
        store_vec8(v_cache + dst, convert_vec8<CT, T>(load_vec8(v_new + src)));
    }
}

// reuse the dispatch macro from elementwise.hip
#define DISPATCH_FLOAT_TYPES_KV(TYPE, NAME, ...)                               \
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

void kv_cache_update(torch::Tensor k_cache, torch::Tensor v_cache,
                     torch::Tensor k_new, torch::Tensor v_new,
                     torch::Tensor positions, torch::Tensor seq_ids) {
    const int B = k_new.size(0), Hkv = k_new.size(1), Sn = k_new.size(2), D = k_new.size(3);
    const int Bc = k_cache.size(0), Smax = k_cache.size(2);
    TORCH_CHECK(D % 8 == 0, "head_dim must be a multiple of 8");
    const bool fp8 = k_cache.scalar_type() == at::ScalarType::Float8_e4m3fn;
    TORCH_CHECK(fp8 || k_cache.scalar_type() == k_new.scalar_type(),
                "cache/new dtype mismatch");
    const long total = (long)B * Hkv * Sn * (D / 8);
    long blocks = (total + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_FLOAT_TYPES_KV(k_new.scalar_type(), "kv_cache_update", [&] {
        if (fp8) {
            hipLaunchKernelGGL((kv_update_kernel<scalar_t, __hip_fp8_e4m3>),
                               dim3((int)blocks), dim3(256), 0, stream,
                               (__hip_fp8_e4m3*)k_cache.data_ptr(),
                               (__hip_fp8_e4m3*)v_cache.data_ptr(),
                               (const scalar_t*)k_new.data_ptr(), (const scalar_t*)v_new.data_ptr(),
                               positions.data_ptr<int>(), seq_ids.data_ptr<int>(),
                               B, Hkv, Sn, D, Bc, Smax);
        } else {
            hipLaunchKernelGGL((kv_update_kernel<scalar_t, scalar_t>),
                               dim3((int)blocks), dim3(256), 0, stream,
                               (scalar_t*)k_cache.data_ptr(), (scalar_t*)v_cache.data_ptr(),
                               (const scalar_t*)k_new.data_ptr(), (const scalar_t*)v_new.data_ptr(),
                               positions.data_ptr<int>(), seq_ids.data_ptr<int>(),
                               B, Hkv, Sn, D, Bc, Smax);
        }
    });
    HIP_CHECK_KERNEL();
}

void block_kv_cache_update(torch::Tensor k_cache, torch::Tensor v_cache,
                           torch::Tensor k_new, torch::Tensor v_new,
                           torch::Tensor slot_mapping) {
    const int N = k_new.size(0), Hkv = k_new.size(1), D = k_new.size(2);
    const long nslots = (long)k_cache.size(0) * k_cache.size(1);
    TORCH_CHECK(D % 8 == 0, "head_dim must be a multiple of 8");
    const long total = (long)N * Hkv * (D / 8);
    long blocks = (total + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    auto stream = at::hip::getCurrentHIPStream();
    const bool fp8 = k_cache.scalar_type() == at::ScalarType::Float8_e4m3fn;
    DISPATCH_FLOAT_TYPES_KV(k_new.scalar_type(), "block_kv_cache_update", [&] {
        if (fp8) {
            hipLaunchKernelGGL((block_kv_update_kernel<scalar_t, __hip_fp8_e4m3>),
                               dim3((int)blocks), dim3(256), 0, stream,
                               (__hip_fp8_e4m3*)k_cache.data_ptr(),
                               (__hip_fp8_e4m3*)v_cache.data_ptr(),
                               (const scalar_t*)k_new.data_ptr(), (const scalar_t*)v_new.data_ptr(),
                               slot_mapping.data_ptr<int>(), N, Hkv, D, nslots);
        } else {
            hipLaunchKernelGGL((block_kv_update_kernel<scalar_t, scalar_t>),
                               dim3((int)blocks), dim3(256), 0, stream,
                               (scalar_t*)k_cache.data_ptr(), (scalar_t*)v_cache.data_ptr(),
                               (const scalar_t*)k_new.data_ptr(), (const scalar_t*)v_new.data_ptr(),
                               slot_mapping.data_ptr<int>(), N, Hkv, D, nslots);
        }
    });
    HIP_CHECK_KERNEL();
}
