// Skinny-M weight-streaming GEMM v3: y = x @ W^T (+bias), M <= 128.
//
// Built from the round-1 TCC counter evidence
// (profiles/r01_skinny_tcc_analysis.md): v2's plateau was (a) atomic
// K-split partials adding 44x the library's EA write traffic and (b) a
// depth-1 prefetch pipeline (one LDS chunk in flight) leaving each WG's
// stream latency-bound at ~4.4 GB/s. v3 removes both:
//
// - NO K-split: each WG register-accumulates its FULL K span and writes
//   bf16 y exactly once (no fp32 workspace, no convert kernel, no RMW).
//   Grid depth comes from the N split x M split instead:
//   (N/32) x ceil(M/64) = 384 WGs for the qkv shape (M=128, N=6144).
// - Depth-3 register prefetch: three staged chunks in flight over a
//   single LDS buffer (~9 outstanding b128 loads per lane vs v2's 3).
// - Weight loads are nontemporal (single-use stream; keeps L2 for x,
//   which every WG re-reads and which stays XCD-L2-resident at 1 MB).
//
// Same LDS pitch/fragment math as v2 (9-slot b128 pitch, conflict-free
// mfma_16x16x32 fragment reads). bf16/f16; N % 32 == 0, K % 64 == 0.
//
// Reference parity: this replaces the reference's fused decode-GEMM
// tiles; the engine wires it only if it beats the
// autotuned hipBLASLt path on the captured shape.

#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __bf16 sk3_bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 sk3_f16x8 __attribute__((ext_vector_type(8)));
typedef float sk3_f32x4 __attribute__((ext_vector_type(4)));
typedef int sk3_i32x4 __attribute__((ext_vector_type(4)));

template <typename T> struct Sk3Mfma;
template <> struct Sk3Mfma<__hip_bfloat16> {
    using vec8 = sk3_bf16x8;
    static __device__ __forceinline__ sk3_f32x4 mfma(vec8 a, vec8 b, sk3_f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    }
    static __device__ __forceinline__ float to_f(__hip_bfloat16 v) {
        return __bfloat162float(v);
    }
    static __device__ __forceinline__ __hip_bfloat16 from_f(float v) {
        return __float2bfloat16(v);
    }
};
template <> struct Sk3Mfma<__half> {
    using vec8 = sk3_f16x8;
    static __device__ __forceinline__ sk3_f32x4 mfma(vec8 a, vec8 b, sk3_f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
    }
    static __device__ __forceinline__ float to_f(__half v) { return __half2float(v); }
    static __device__ __forceinline__ __half from_f(float v) { return __float2half(v); }
};

constexpr int KC3 = 64;                // K elems per LDS chunk
constexpr int PITCH3 = KC3 + 8;        // 9 b128 slots: conflict-free frags
constexpr int NT3 = 64;                // W rows per WG (x logical traffic
                                       // scales as N/NT3: PMC showed the
                                       // x re-read was the 1.56x over-fetch)
constexpr int MT3 = 64;                // x rows per WG (4 waves x 16)

// 4 waves each own one 16-row m-tile and loop the 4 n-tiles.
template <typename T>
__launch_bounds__(256, 4)
__global__ void skinny_gemm3_kernel(
    T* __restrict__ y,            // (M, N)
    const T* __restrict__ x,      // (M, K)
    const T* __restrict__ w,      // (N, K)
    const T* __restrict__ bias,   // (N,) or nullptr
    const int M, const int N, const int K) {
    using MF = Sk3Mfma<T>;
    using vec8 = typename MF::vec8;

    const int n_base = blockIdx.x * NT3;
    const int m_base = blockIdx.y * MT3;
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int col = lane & 15;
    const int kseg = lane >> 4;

    __shared__ T wt[NT3 * PITCH3];
    __shared__ T xt[MT3 * PITCH3];

    // staging maps (one chunk = W 32x64 = 4 KB + x 64x64 = 8 KB):
    // W: 8 threads per 128 B row -> 1 x b128 per thread
    const int ww_row = threadIdx.x >> 2;
    const int ww_off = (threadIdx.x & 3) * 16;  // elems
    // x: 4 threads per 128 B row -> 2 x b128 per thread
    const int xx_row = threadIdx.x >> 2;
    const int xx_off = (threadIdx.x & 3) * 16;  // elems

    // SPLIT prefetch queues (the single-queue version collapsed: 6 fused
    // stages x 12 VGPRs + temps exceeded the 128-VGPR budget and the
    // allocator reused stage registers, shrinking the pipeline to depth
    // ~1.5 — the ISA showed `s_waitcnt vmcnt(2)`/`vmcnt(0)` at every
    // commit). W is the HBM stream and needs the deep queue (8 stages x
    // 4 VGPRs); x is XCD-L2-resident (192 WGs re-read the same 1 MB) so
    // a short 4-stage queue covers its latency.
    auto load_w = [&](int k0, sk3_i32x4* r) {
#pragma unroll
        for (int p = 0; p < 2; ++p)
            r[p] = __builtin_nontemporal_load(reinterpret_cast<const sk3_i32x4*>(
                w + (long)(n_base + ww_row) * K + k0 + ww_off + p * 8));
    };
    auto load_x = [&](int k0, sk3_i32x4* r) {
        const int xrow = min(m_base + xx_row, M - 1);
#pragma unroll
        for (int p = 0; p < 2; ++p)
            r[p] = *reinterpret_cast<const sk3_i32x4*>(
                x + (long)xrow * K + k0 + xx_off + p * 8);
    };
    auto commit_w = [&](const sk3_i32x4* r) {
#pragma unroll
        for (int p = 0; p < 2; ++p)
            *reinterpret_cast<sk3_i32x4*>(wt + ww_row * PITCH3 + ww_off + p * 8) = r[p];
    };
    auto commit_x = [&](const sk3_i32x4* r) {
#pragma unroll
        for (int p = 0; p < 2; ++p)
            *reinterpret_cast<sk3_i32x4*>(xt + xx_row * PITCH3 + xx_off + p * 8) = r[p];
    };

    sk3_f32x4 acc[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[n] = sk3_f32x4{0.f, 0.f, 0.f, 0.f};

    const int nchunks = K / KC3;
    // NAMED stages, modulo schedule unrolled x6 (W queue 6 x 2 b128,
    // x queue 3 x 2 b128): dynamic indexing of a stage array would
    // demote it to scratch (160 B/lane measured)
    sk3_i32x4 w0[2], w1[2], w2[2], w3[2], w4[2], w5[2];
    sk3_i32x4 x0[2], x1[2], x2[2];
    load_w(0, w0); load_w(KC3, w1); load_w(2 * KC3, w2); load_w(3 * KC3, w3);
    load_w(4 * KC3, w4); load_w(5 * KC3, w5);
    load_x(0, x0); load_x(KC3, x1); load_x(2 * KC3, x2);
    commit_w(w0);
    commit_x(x0);

    auto iter_body = [&](int i, sk3_i32x4* w_reuse, sk3_i32x4* w_next,
                         sk3_i32x4* x_reuse, sk3_i32x4* x_next, bool steady) {
        // consume chunk i from LDS; commit chunk i+1; refill the stages
        // that held chunk i with chunks i+6 (W) and i+3 (x). In the
        // steady state everything is UNCONDITIONAL (loads clamped): a
        // load behind a branch breaks the compiler's static outstanding-
        // load count and every later ds_write gets `s_waitcnt vmcnt(0)`,
        // draining the whole queue (measured: no gain from depth 3->6
        // until the branches went away).
        __syncthreads();                 // chunk i visible in LDS
#pragma unroll
        for (int ks = 0; ks < KC3 / 32; ++ks) {
            const vec8 afrag = *reinterpret_cast<const vec8*>(
                xt + (wave * 16 + col) * PITCH3 + ks * 32 + kseg * 8);
#pragma unroll
            for (int n = 0; n < 4; ++n) {
                const vec8 bfrag = *reinterpret_cast<const vec8*>(
                    wt + (n * 16 + col) * PITCH3 + ks * 32 + kseg * 8);
                acc[n] = MF::mfma(afrag, bfrag, acc[n]);
            }
        }
        __syncthreads();                 // LDS reads done before restage
        if (steady) {
            commit_w(w_next);            // chunk i+1 -> LDS
            commit_x(x_next);
            load_w(min(i + 6, nchunks - 1) * KC3, w_reuse);
            load_x(min(i + 3, nchunks - 1) * KC3, x_reuse);
            // pin the schedule: without this the scheduler sinks stage
            // loads toward their commits, collapsing the latency cover
            __builtin_amdgcn_sched_barrier(0);
        } else if (i + 1 < nchunks) {
            commit_w(w_next);
            commit_x(x_next);
            // the x queue is only 3 deep vs the 6-body tail: the first
            // tail bodies must still refill it (W's 6-deep queue is fully
            // loaded by the steady loop)
            if (i + 3 < nchunks) load_x((i + 3) * KC3, x_reuse);
        }
    };
    int i = 0;
    for (; i + 6 < nchunks; i += 6) {
        iter_body(i + 0, w0, w1, x0, x1, true);
        iter_body(i + 1, w1, w2, x1, x2, true);
        iter_body(i + 2, w2, w3, x2, x0, true);
        iter_body(i + 3, w3, w4, x0, x1, true);
        iter_body(i + 4, w4, w5, x1, x2, true);
        iter_body(i + 5, w5, w0, x2, x0, true);
    }
    // tail: every remaining W chunk is already loaded; x refills guarded
    {
        iter_body(i + 0, w0, w1, x0, x1, false);
        if (i + 1 < nchunks) iter_body(i + 1, w1, w2, x1, x2, false);
        if (i + 2 < nchunks) iter_body(i + 2, w2, w3, x2, x0, false);
        if (i + 3 < nchunks) iter_body(i + 3, w3, w4, x0, x1, false);
        if (i + 4 < nchunks) iter_body(i + 4, w4, w5, x1, x2, false);
        if (i + 5 < nchunks) iter_body(i + 5, w5, w0, x2, x0, false);
    }

#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int m = m_base + wave * 16 + kseg * 4 + r;
            if (m >= M) continue;
            const int nn = n_base + n * 16 + col;
            float v = acc[n][r];
            if (bias != nullptr) v += MF::to_f(bias[nn]);
            y[(long)m * N + nn] = MF::from_f(v);
        }
    }
}

template <typename T>
void launch_sk3(torch::Tensor& y, const torch::Tensor& x, const torch::Tensor& w,
                const torch::Tensor* bias) {
    const int M = x.size(0), K = x.size(1), N = w.size(0);
    auto stream = at::hip::getCurrentHIPStream();
    dim3 grid(N / NT3, (M + MT3 - 1) / MT3);
    hipLaunchKernelGGL((skinny_gemm3_kernel<T>), grid, dim3(256), 0, stream,
                       (T*)y.data_ptr(), (const T*)x.data_ptr(),
                       (const T*)w.data_ptr(),
                       bias ? (const T*)bias->data_ptr() : nullptr, M, N, K);
    HIP_CHECK_KERNEL();
}

}  // namespace

void skinny_gemm3(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                  c10::optional<torch::Tensor> bias_opt) {
    const int M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(M <= 1024 && N % NT3 == 0 && K % KC3 == 0 && K >= 6 * KC3,
                "skinny_gemm3: M<=1024, N%64, K%64, K>=384");
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && y.is_contiguous());
    torch::Tensor bias_t;
    const torch::Tensor* bias = nullptr;
    if (bias_opt.has_value()) {
        bias_t = bias_opt->contiguous();
        bias = &bias_t;
    }
    if (x.scalar_type() == at::ScalarType::BFloat16) {
        launch_sk3<__hip_bfloat16>(y, x, w, bias);
    } else if (x.scalar_type() == at::ScalarType::Half) {
        launch_sk3<__half>(y, x, w, bias);
    } else {
        TORCH_CHECK(false, "skinny_gemm3: bf16/f16 only");
    }
}
