// EXPERIMENTAL blocked skinny-M GEMM v2: y = x @ W^T, M <= 128.
//
// Status (round 1, measured cold-weight): 0.78-1.12 TB/s vs 1.4-2.3 for
// the autotuned hipBLASLt path — NOT wired into the hot path. v2 fixed
// v1's intensity bug (x now re-read only N/128 times) and the occupancy
// (KC=64, 37 KB LDS, 4 WGs/CU), yet both hand kernels plateau near
// 1 TB/s while the library reaches 2+ and a raw contiguous read ~3 on
// the same footprints. The unexplained residual needs a TCC/TCP counter
// deep-dive (round 2); the engine keeps hipBLASLt+timed-autotune for the
// decode GEMMs, which is within ~1.5x of the raw-read ceiling.
//
// Design (kept for round 2):
//
// - 128 W rows per workgroup (x re-read only N/128 times, and x stays
//   MALL-resident), both x and W staged through LDS (32 KB + 32 KB,
//   17-slot padded pitch -> conflict-free fragment reads);
// - K-split across workgroups for grid depth (>= ~512 WGs); each WG
//   accumulates its K-span in registers and commits ONCE per output via
//   global fp32 atomics into a (M, N) fp32 accumulator that fits in
//   L2/MALL (3 MB for qkv), so the split costs no HBM round-trip of
//   partial tiles;
// - 4 waves split the 128 columns (2 n-tiles each); mfma_16x16x32;
//   register-prefetched chunks with deferred LDS commit (the skinny v1
//   lesson: never let the wave block on the HBM round-trip before the
//   MFMAs).
//
// The fp32 accumulator is zeroed by the caller and converted to bf16 by
// a trailing elementwise kernel. bf16/f16; K % 128 == 0, N % 128 == 0.

#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __bf16 sk2_bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 sk2_f16x8 __attribute__((ext_vector_type(8)));
typedef float sk2_f32x4 __attribute__((ext_vector_type(4)));

template <typename T> struct Sk2Mfma;
template <> struct Sk2Mfma<__hip_bfloat16> {
    using vec8 = sk2_bf16x8;
    static __device__ __forceinline__ sk2_f32x4 mfma(vec8 a, vec8 b, sk2_f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    }
};
template <> struct Sk2Mfma<__half> {
    using vec8 = sk2_f16x8;
    static __device__ __forceinline__ sk2_f32x4 mfma(vec8 a, vec8 b, sk2_f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
    }
};

constexpr int KC = 64;               // K elems per LDS chunk
constexpr int PITCH = KC + 8;        // 9 b128 slots: conflict-free frags
constexpr int NT = 128;              // W rows per WG

// MT = ceil(M/16) m-tiles; 4 waves each own 2 n-tiles of 16 cols.
template <typename T, int MT>
__launch_bounds__(256, 4)
__global__ void skinny_gemm2_kernel(
    float* __restrict__ y32,      // (M, N) fp32 accumulator (pre-zeroed)
    const T* __restrict__ x,      // (M, K)
    const T* __restrict__ w,      // (N, K)
    const int M, const int N, const int K, const int KQ) {
    using MF = Sk2Mfma<T>;
    using vec8 = typename MF::vec8;

    const int n_base = blockIdx.x * NT;
    const int k_lo = blockIdx.y * KQ;
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int col = lane & 15;
    const int kseg = lane >> 4;

    __shared__ T wt[NT * PITCH];
    __shared__ T xt[128 * PITCH];

    // staging: thread t covers row t>>1, half (t&1) of a 256B row
    const int st_row = threadIdx.x >> 1;
    const int st_off = (threadIdx.x & 1) * 32;  // elems (half row)

    auto load_chunk = [&](int k0, int4* wr, int4* xr) {
#pragma unroll
        for (int p = 0; p < 4; ++p)
            wr[p] = *reinterpret_cast<const int4*>(
                w + (long)(n_base + st_row) * K + k0 + st_off + p * 8);
        const int xrow = min(st_row, M - 1);
#pragma unroll
        for (int p = 0; p < 4; ++p)
            xr[p] = *reinterpret_cast<const int4*>(
                x + (long)xrow * K + k0 + st_off + p * 8);
    };
    auto commit_chunk = [&](const int4* wr, const int4* xr) {
#pragma unroll
        for (int p = 0; p < 4; ++p)
            *reinterpret_cast<int4*>(wt + st_row * PITCH + st_off + p * 8) = wr[p];
#pragma unroll
        for (int p = 0; p < 4; ++p)
            *reinterpret_cast<int4*>(xt + st_row * PITCH + st_off + p * 8) = xr[p];
    };

    sk2_f32x4 acc[MT][2];
#pragma unroll
    for (int t = 0; t < MT; ++t)
#pragma unroll
        for (int n = 0; n < 2; ++n) acc[t][n] = sk2_f32x4{0.f, 0.f, 0.f, 0.f};

    int4 wreg[8], xreg[8];
    load_chunk(k_lo, wreg, xreg);
    commit_chunk(wreg, xreg);
    const int k_hi = k_lo + KQ;
    for (int k0 = k_lo; k0 < k_hi; k0 += KC) {
        __syncthreads();                 // LDS chunk visible
        if (k0 + KC < k_hi) load_chunk(k0 + KC, wreg, xreg);
#pragma unroll
        for (int ks = 0; ks < KC / 32; ++ks) {
            vec8 bfrag[2];
#pragma unroll
            for (int n = 0; n < 2; ++n)
                bfrag[n] = *reinterpret_cast<const vec8*>(
                    wt + (wave * 32 + n * 16 + col) * PITCH + ks * 32 + kseg * 8);
#pragma unroll
            for (int t = 0; t < MT; ++t) {
                const vec8 afrag = *reinterpret_cast<const vec8*>(
                    xt + (t * 16 + col) * PITCH + ks * 32 + kseg * 8);
#pragma unroll
                for (int n = 0; n < 2; ++n)
                    acc[t][n] = MF::mfma(afrag, bfrag[n], acc[t][n]);
            }
        }
        __syncthreads();                 // reads done before restage
        if (k0 + KC < k_hi) commit_chunk(wreg, xreg);
    }

    // one fp32 atomic per output element per WG (accumulator is
    // MALL-resident; gridDim.y WGs contribute per element)
#pragma unroll
    for (int t = 0; t < MT; ++t) {
#pragma unroll
        for (int n = 0; n < 2; ++n) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = t * 16 + kseg * 4 + r;
                if (m >= M) continue;
                const long o = (long)m * N + n_base + wave * 32 + n * 16 + col;
                if (gridDim.y == 1) {
                    y32[o] = acc[t][n][r];
                } else {
                    atomicAdd(y32 + o, acc[t][n][r]);
                }
            }
        }
    }
}

template <typename T>
__global__ void sk2_convert_kernel(T* __restrict__ y, const float* __restrict__ y32,
                                   const T* __restrict__ bias, const long total,
                                   const int N) {
    const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < total) {
        float v = y32[i];
        if (bias != nullptr) v += VecTraits<T>::to_float(bias[i % N]);
        y[i] = VecTraits<T>::from_float(v);
    }
}

template <typename T>
void launch_sk2(torch::Tensor& y, torch::Tensor& y32, const torch::Tensor& x,
                const torch::Tensor& w, const torch::Tensor* bias, int ksplit) {
    const int M = x.size(0), K = x.size(1), N = w.size(0);
    auto stream = at::hip::getCurrentHIPStream();
    const int KQ = K / ksplit;
    dim3 grid(N / NT, ksplit);
#define SK2_CASE(MT)                                                          \
    hipLaunchKernelGGL((skinny_gemm2_kernel<T, MT>), grid, dim3(256), 0,      \
                       stream, y32.data_ptr<float>(), (const T*)x.data_ptr(), \
                       (const T*)w.data_ptr(), M, N, K, KQ)
    switch ((M + 15) / 16) {
        case 1: SK2_CASE(1); break;
        case 2: SK2_CASE(2); break;
        case 3: SK2_CASE(3); break;
        case 4: SK2_CASE(4); break;
        case 5: SK2_CASE(5); break;
        case 6: SK2_CASE(6); break;
        case 7: SK2_CASE(7); break;
        case 8: SK2_CASE(8); break;
        default: TORCH_CHECK(false, "skinny_gemm2: M > 128");
    }
#undef SK2_CASE
    HIP_CHECK_KERNEL();
    const long total = (long)M * N;
    const int thr = 256;
    hipLaunchKernelGGL((sk2_convert_kernel<T>), dim3((total + thr - 1) / thr),
                       dim3(thr), 0, stream, (T*)y.data_ptr(),
                       y32.data_ptr<float>(),
                       bias ? (const T*)bias->data_ptr() : nullptr, total, N);
    HIP_CHECK_KERNEL();
}

}  // namespace

void skinny_gemm2(torch::Tensor y, torch::Tensor y32, torch::Tensor x,
                  torch::Tensor w, c10::optional<torch::Tensor> bias_opt,
                  long ksplit) {
    const int M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(M <= 128 && N % NT == 0 && K % (KC * ksplit) == 0,
                "skinny_gemm2: M<=128, N%128, K%(KC*ksplit)");
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && y.is_contiguous());
    TORCH_CHECK(y32.scalar_type() == torch::kFloat32 &&
                y32.numel() == (long)M * N);
    torch::Tensor bias_t;
    const torch::Tensor* bias = nullptr;
    if (bias_opt.has_value()) {
        bias_t = bias_opt->contiguous();
        bias = &bias_t;
    }
    if (x.scalar_type() == at::ScalarType::BFloat16) {
        launch_sk2<__hip_bfloat16>(y, y32, x, w, bias, (int)ksplit);
    } else if (x.scalar_type() == at::ScalarType::Half) {
        launch_sk2<__half>(y, y32, x, w, bias, (int)ksplit);
    } else {
        TORCH_CHECK(false, "skinny_gemm2: bf16/f16 only");
    }
}
