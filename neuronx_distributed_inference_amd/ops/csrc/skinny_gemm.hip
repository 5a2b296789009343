// EXPERIMENTAL skinny-M GEMM: y(M,N) = x(M,K) @ W(N,K)^T [+ bias], M <= 128.
//
// Status (round 1, measured cold-weight on MI355X, scripts/perf_skinny.py):
// this kernel reaches ~0.7-1.1 TB/s of W traffic vs 1.4-4.6 TB/s for the
// autotuned hipBLASLt path, and is NOT wired into the hot path. The
// measured story, kept for round 2:
//   - per-WG structure (LDS-staged W, register-prefetched A fragments,
//     conflict-free LDS pitch, deferred LDS commit) is latency-clean: 8
//     waves/SIMD, no spills, zero-conflict B reads;
//   - the binding constraint is ARITHMETIC INTENSITY of the tile shape:
//     with 16-column N tiles, x (M x K) is re-read by every WG, i.e.
//     N/16 times -> x traffic = (M/16) x W traffic = 8x at M=128. The
//     padded-lda experiment ruled out HBM channel conflicts.
//   - the fix is a full blocked GEMM (NT >= 128 with x and W both staged
//     in LDS + K-split for small-N shapes) — i.e. out-engineering the
//     library tile; hipBLASLt's timed-autotune path (lt_gemm.cpp) already
//     gets 2.0-4.6 TB/s on the decode shapes, so that rewrite is a
//     round-2 item (GAPS.md).
//
// The kernel itself is numerically exact vs F.linear and covered by a GPU
// test; fp16 uses the f16 MFMA form. K % 256 == 0, N % 16 == 0, lda
// (padded W row pitch) supported.

#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef _Float16 f16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

template <typename T> struct SkMfma;
template <> struct SkMfma<__hip_bfloat16> {
    using vec8 = bf16x8_t;
    static __device__ __forceinline__ f32x4_t mfma(vec8 a, vec8 b, f32x4_t c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    }
};
template <> struct SkMfma<__half> {
    using vec8 = f16x8_t;
    static __device__ __forceinline__ f32x4_t mfma(vec8 a, vec8 b, f32x4_t c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
    }
};

constexpr int KC = 256;                    // K elems per LDS chunk
// pitch 36 b128 slots: B-frag lane (col,kseg) hits slot 36*col+kseg whose
// residues mod 64 are exactly the 64 distinct values - zero bank conflicts
constexpr int LDS_PITCH = 288;

// MT = total 16-row M tiles (ceil(M/16)); waves split them.
template <typename T, int MT>
__launch_bounds__(256, 4)
__global__ void skinny_gemm_kernel(
    T* __restrict__ y,            // (M, N)
    const T* __restrict__ x,      // (M, K)
    const T* __restrict__ w,      // (N, K)
    const T* __restrict__ bias,   // (N,) or null
    const int M, const int N, const int K,
    const int lda) {              // W row pitch in elems (>= K)
    using MF = SkMfma<T>;
    using vec8 = typename MF::vec8;

    const int n_base = blockIdx.x * 16;
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int col = lane & 15;             // fragment col (B: n, A: m-row)
    const int kseg = lane >> 4;            // 8-elem K slice of the fragment

    __shared__ T wtile[2][16 * LDS_PITCH];

    // W chunk staged via REGISTERS: loads for chunk i+1 are issued before
    // chunk i's MFMAs and only committed to LDS afterwards — the ds_write
    // is where the wave waits on vmcnt, so issuing it after the compute
    // hides the whole HBM round-trip behind the MFMAs (a stage that
    // loads+writes up front serializes: the wave blocks at the write).
    const int st_row = threadIdx.x >> 4;
    const int st_off = (threadIdx.x & 15) * 8;  // elems

    auto load_w = [&](int k0, int4* reg) {
#pragma unroll
        for (int p = 0; p < 2; ++p)
            reg[p] = *reinterpret_cast<const int4*>(
                w + (long)(n_base + st_row) * lda + k0 + st_off + p * 128);
    };
    auto commit_w = [&](int buf, const int4* reg) {
#pragma unroll
        for (int p = 0; p < 2; ++p)
            *reinterpret_cast<int4*>(
                wtile[buf] + st_row * LDS_PITCH + st_off + p * 128) = reg[p];
    };

    // this wave's M tiles: wave, wave+4, ... (round-robin over MT)
    constexpr int TPW = (MT + 3) / 4;  // tiles per wave (upper bound)
    f32x4_t acc[TPW];
#pragma unroll
    for (int i = 0; i < TPW; ++i) acc[i] = f32x4_t{0.f, 0.f, 0.f, 0.f};

    int4 wreg[2];
    load_w(0, wreg);
    commit_w(0, wreg);
    for (int k0 = 0; k0 < K; k0 += KC) {
        const int buf = (k0 / KC) & 1;
        __syncthreads();                       // tile visible to all waves
        // issue next chunk's W loads first: they stay in flight across the
        // A-loads and MFMAs below
        if (k0 + KC < K) load_w(k0 + KC, wreg);
        vec8 areg[TPW][KC / 32];
#pragma unroll
        for (int i = 0; i < TPW; ++i) {
            const int t = wave + i * 4;
            if (t >= MT) break;
            const int row = min(t * 16 + col, M - 1);  // clamped pad rows
#pragma unroll
            for (int ks = 0; ks < KC / 32; ++ks)
                areg[i][ks] = *reinterpret_cast<const vec8*>(
                    x + (long)row * K + k0 + ks * 32 + kseg * 8);
        }
#pragma unroll
        for (int ks = 0; ks < KC / 32; ++ks) {
            const vec8 bfrag = *reinterpret_cast<const vec8*>(
                wtile[buf] + col * LDS_PITCH + ks * 32 + kseg * 8);
#pragma unroll
            for (int i = 0; i < TPW; ++i) {
                if (wave + i * 4 >= MT) break;
                acc[i] = MF::mfma(areg[i][ks], bfrag, acc[i]);
            }
        }
        if (k0 + KC < K) commit_w(buf ^ 1, wreg);
        __syncthreads();  // commits done before the next read of buf^1
    }

    // epilogue: C frag row = (lane>>4)*4 + reg, col = lane&15
#pragma unroll
    for (int i = 0; i < TPW; ++i) {
        const int t = wave + i * 4;
        if (t >= MT) break;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int m = t * 16 + kseg * 4 + r;
            if (m >= M) continue;
            float v = acc[i][r];
            if (bias != nullptr) v += VecTraits<T>::to_float(bias[n_base + col]);
            y[(long)m * N + n_base + col] = VecTraits<T>::from_float(v);
        }
    }
}

template <typename T>
void launch_skinny(torch::Tensor& y, const torch::Tensor& x,
                   const torch::Tensor& w, const torch::Tensor* bias) {
    const int M = x.size(0), K = x.size(1), N = w.size(0);
    const int lda = w.stride(0);
    auto stream = at::hip::getCurrentHIPStream();
    const T* bp = bias ? (const T*)bias->data_ptr() : nullptr;
    dim3 grid(N / 16);
#define MT_CASE(MT)                                                           \
    hipLaunchKernelGGL((skinny_gemm_kernel<T, MT>), grid, dim3(256), 0,       \
                       stream, (T*)y.data_ptr(), (const T*)x.data_ptr(),      \
                       (const T*)w.data_ptr(), bp, M, N, K, lda)
    const int mt = (M + 15) / 16;
    switch (mt) {
        case 1: MT_CASE(1); break;
        case 2: MT_CASE(2); break;
        case 3: MT_CASE(3); break;
        case 4: MT_CASE(4); break;
        case 5: MT_CASE(5); break;
        case 6: MT_CASE(6); break;
        case 7: MT_CASE(7); break;
        case 8: MT_CASE(8); break;
        default: TORCH_CHECK(false, "skinny_gemm: M > 128");
    }
#undef MT_CASE
    HIP_CHECK_KERNEL();
}

}  // namespace

void skinny_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                 c10::optional<torch::Tensor> bias_opt) {
    TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && y.dim() == 2, "2D only");
    const int M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(M <= 128, "skinny_gemm: M <= 128 (got ", M, ")");
    TORCH_CHECK(K % 256 == 0 && N % 16 == 0, "skinny_gemm: K%256, N%16");
    TORCH_CHECK(x.is_contiguous() && y.is_contiguous());
    TORCH_CHECK(w.stride(1) == 1 && w.stride(0) >= K,
                "W must be row-major (padded row pitch allowed)");
    TORCH_CHECK(w.size(1) == K && y.size(0) == M && y.size(1) == N);
    torch::Tensor bias_t;
    const torch::Tensor* bias = nullptr;
    if (bias_opt.has_value()) {
        bias_t = bias_opt->contiguous();
        bias = &bias_t;
    }
    if (x.scalar_type() == at::ScalarType::BFloat16) {
        launch_skinny<__hip_bfloat16>(y, x, w, bias);
    } else if (x.scalar_type() == at::ScalarType::Half) {
        launch_skinny<__half>(y, x, w, bias);
    } else {
        TORCH_CHECK(false, "skinny_gemm: bf16/f16 only");
    }
}
