// Common helpers for the CDNA4 (gfx950) kernels.
// Wavefront = 64 lanes; LDS 160 KiB/CU; bf16 loads vectorized as short4/short8
// per the CDNA HIP guide (G13: hipcc does not auto-vectorize bf16 loads).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_fp8.h>

#define WAVE_SIZE 64

#define DEV_INLINE __device__ __forceinline__

// ------------------------- scalar conversion traits -------------------------
template <typename T> struct VecTraits;

template <> struct VecTraits<__hip_bfloat16> {
    using scalar_t = __hip_bfloat16;
    static DEV_INLINE float to_float(scalar_t v) { return __bfloat162float(v); }
    static DEV_INLINE scalar_t from_float(float f) { return __float2bfloat16(f); }
};
template <> struct VecTraits<__half> {
    using scalar_t = __half;
    static DEV_INLINE float to_float(scalar_t v) { return __half2float(v); }
    static DEV_INLINE scalar_t from_float(float f) { return __float2half(f); }
};
template <> struct VecTraits<float> {
    using scalar_t = float;
    static DEV_INLINE float to_float(scalar_t v) { return v; }
    static DEV_INLINE scalar_t from_float(float f) { return f; }
};
// OCP FP8 e4m3fn (gfx950-native format — NOT the MI300X fnuz variant)
template <> struct VecTraits<__hip_fp8_e4m3> {
    using scalar_t = __hip_fp8_e4m3;
    static DEV_INLINE float to_float(scalar_t v) { return (float)v; }
    static DEV_INLINE scalar_t from_float(float f) { return scalar_t(f); }
};

// 8-element vector of T loaded as one 16B (bf16/f16) or 32B (f32: two 16B) op.
template <typename T>
struct Vec8 {
    T v[8];
};

template <typename T>
DEV_INLINE Vec8<T> load_vec8(const T* p) {
    Vec8<T> out;
    if constexpr (sizeof(T) == 1) {
        *reinterpret_cast<int2*>(out.v) = *reinterpret_cast<const int2*>(p);
    } else if constexpr (sizeof(T) == 2) {
        *reinterpret_cast<int4*>(out.v) = *reinterpret_cast<const int4*>(p);
    } else {
        *reinterpret_cast<int4*>(out.v) = *reinterpret_cast<const int4*>(p);
        *reinterpret_cast<int4*>(out.v + 4) = *reinterpret_cast<const int4*>(p + 4);
    }
    return out;
}

template <typename T>
DEV_INLINE void store_vec8(T* p, const Vec8<T>& x) {
    if constexpr (sizeof(T) == 1) {
        *reinterpret_cast<int2*>(p) = *reinterpret_cast<const int2*>(x.v);
    } else if constexpr (sizeof(T) == 2) {
        *reinterpret_cast<int4*>(p) = *reinterpret_cast<const int4*>(x.v);
    } else {
        *reinterpret_cast<int4*>(p) = *reinterpret_cast<const int4*>(x.v);
        *reinterpret_cast<int4*>(p + 4) = *reinterpret_cast<const int4*>(x.v + 4);
    }
}

// convert an 8-vector between storage types through fp32
template <typename DST, typename SRC>
DEV_INLINE Vec8<DST> convert_vec8(const Vec8<SRC>& x, float scale = 1.0f) {
    Vec8<DST> out;
#pragma unroll
    for (int j = 0; j < 8; ++j)
        out.v[j] = VecTraits<DST>::from_float(VecTraits<SRC>::to_float(x.v[j]) * scale);
    return out;
}

// ------------------------- wave reductions -------------------------
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
    return v;
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
    return v;
}

// reduce within a contiguous lane group of size GS (power of two <= 64)
template <int GS>
DEV_INLINE float group_reduce_sum(float v) {
#pragma unroll
    for (int off = GS / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
    return v;
}

// block reduction across waves through LDS (max 16 waves)
DEV_INLINE float block_reduce_sum(float v, float* lds_scratch) {
    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wid = threadIdx.x / WAVE_SIZE;
    const int nwaves = blockDim.x / WAVE_SIZE;
    v = wave_reduce_sum(v);
    if (lane == 0) lds_scratch[wid] = v;
    __syncthreads();
    float out = (threadIdx.x < nwaves) ? lds_scratch[threadIdx.x] : 0.f;
    if (wid == 0) {
        out = wave_reduce_sum(out);
        if (lane == 0) lds_scratch[0] = out;
    }
    __syncthreads();
    float r = lds_scratch[0];
    __syncthreads();
    return r;
}

DEV_INLINE int cdiv(int a, int b) { return (a + b - 1) / b; }

#define HIP_CHECK_KERNEL()                                                     \
    do {                                                                       \
        hipError_t e = hipGetLastError();                                      \
        TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",             \
                    hipGetErrorString(e));                                     \
    } while (0)
