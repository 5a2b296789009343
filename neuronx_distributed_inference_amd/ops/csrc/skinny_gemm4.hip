// Skinny-M weight-streaming GEMM v4: y = x @ W^T (+bias), M <= 1024.
//
// STATUS: EXPERIMENTAL / NOT SAFE TO CALL — faults on gfx950 (memory
// fault, core dump). Root cause class: the compiler treats inline-asm
// load destinations as written at the asm point, so it may reuse an
// in-flight destination register (or a pair overlapping one) before
// the hand-placed s_waitcnt lands the data. The correct round-2 shape
// is the guide's `global_load_lds` template: no destination VGPRs at
// all (data lands in LDS), raw s_barrier with lgkmcnt-only waits, and
// counted vmcnt drains at the read points — which also requires a
// LINEAR (unpitched) LDS image with an XOR source-swizzle to keep the
// fragment reads conflict-free. Kept as the starting point + wait-count
// math for that kernel; no python test invokes it.
//
// v3 established (profiles/r01_skinny_tcc_analysis.md) that every
// compiler-scheduled variant pays ~latency/3 of stall per K-chunk:
// hipcc will not hold more than ~3 bodies of loads outstanding across
// an LDS commit, independent of structure. v4 removes the compiler from
// the vmem pipeline entirely:
//
// - ALL global loads in the hot loop are inline-asm
//   `global_load_dwordx4` (nt on the W stream), so no compiler waitcnt
//   insertion happens for them;
// - the ONLY waits are hand-counted `s_waitcnt vmcnt(N)` before each
//   LDS commit, with N derived from the static issue schedule (see the
//   table below) — the full 5-body W queue becomes real latency cover;
// - LDS visibility across waves still uses __syncthreads() (hipcc emits
//   lgkmcnt-only waits at barriers when no tracked vmem is pending),
//   which also closes the stage-register WAR window: a stage reloaded
//   at body i was ds-committed at body i-1, and the barrier between
//   them drains lgkm.
//
// Static wait counts (loads per body = 2 W + 2 x, issued W-then-x
// after the commits; BOTH queues are 6 deep so chunk i+1's loads were
// issued at body i-5): loads issued after W(i+1) = its body's x pair
// (2) + 4 bodies x 4 = 18; after x(i+1) = 16. One full drain
// (vmcnt(0)) right after the prologue makes those steady counts safe
// for the first bodies too (their chunks are prologue-landed, and by
// body 5 the issue counts have reached steady state). A shallower x
// queue does NOT work: vmcnt is one ordered counter, so waiting for a
// young x load drains the older W queue with it (measured as the v3
// collapse).
//
// Tiling identical to v3: NT=64 W rows x MT=64 x rows per WG, 4 waves
// each own one 16-row m-tile and loop 4 n-tiles; KC=64 chunks staged
// through LDS with a 9-slot b128 pitch; W and x queues 6 deep.
// bf16/f16; N % 64 == 0, K % 64 == 0, K >= 384.

#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __bf16 sk4_bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 sk4_f16x8 __attribute__((ext_vector_type(8)));
typedef float sk4_f32x4 __attribute__((ext_vector_type(4)));
typedef int sk4_i32x4 __attribute__((ext_vector_type(4)));

template <typename T> struct Sk4Mfma;
template <> struct Sk4Mfma<__hip_bfloat16> {
    using vec8 = sk4_bf16x8;
    static __device__ __forceinline__ sk4_f32x4 mfma(vec8 a, vec8 b, sk4_f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    }
    static __device__ __forceinline__ float to_f(__hip_bfloat16 v) {
        return __bfloat162float(v);
    }
    static __device__ __forceinline__ __hip_bfloat16 from_f(float v) {
        return __float2bfloat16(v);
    }
};
template <> struct Sk4Mfma<__half> {
    using vec8 = sk4_f16x8;
    static __device__ __forceinline__ sk4_f32x4 mfma(vec8 a, vec8 b, sk4_f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
    }
    static __device__ __forceinline__ float to_f(__half v) { return __half2float(v); }
    static __device__ __forceinline__ __half from_f(float v) { return __float2half(v); }
};

constexpr int KC4 = 64;
constexpr int PITCH4 = KC4 + 8;
constexpr int NT4 = 64;
constexpr int MT4 = 64;

__device__ __forceinline__ sk4_i32x4 glb_load_nt(const void* p) {
    sk4_i32x4 r;
    asm volatile("global_load_dwordx4 %0, %1, off nt"
                 : "=v"(r) : "v"(p));
    return r;
}
__device__ __forceinline__ sk4_i32x4 glb_load(const void* p) {
    sk4_i32x4 r;
    asm volatile("global_load_dwordx4 %0, %1, off"
                 : "=v"(r) : "v"(p));
    return r;
}
#define SK4_WAIT(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")

template <typename T>
__launch_bounds__(256, 3)
__global__ void skinny_gemm4_kernel(
    T* __restrict__ y, const T* __restrict__ x, const T* __restrict__ w,
    const T* __restrict__ bias, const int M, const int N, const int K) {
    using MF = Sk4Mfma<T>;
    using vec8 = typename MF::vec8;

    const int n_base = blockIdx.x * NT4;
    const int m_base = blockIdx.y * MT4;
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int col = lane & 15;
    const int kseg = lane >> 4;

    __shared__ T wt[NT4 * PITCH4];
    __shared__ T xt[MT4 * PITCH4];

    // staging: 4 threads per 128 B row -> 2 b128 per thread, both arrays
    const int st_row = threadIdx.x >> 2;
    const int st_off = (threadIdx.x & 3) * 16;  // elems

    const T* wbase = w + (long)(n_base + st_row) * K + st_off;
    const int xrow = min(m_base + st_row, M - 1);
    const T* xbase = x + (long)xrow * K + st_off;

    auto load_w = [&](int k0, sk4_i32x4* r) {
        r[0] = glb_load_nt(wbase + k0);
        r[1] = glb_load_nt(wbase + k0 + 8);
    };
    auto load_x = [&](int k0, sk4_i32x4* r) {
        r[0] = glb_load(xbase + k0);
        r[1] = glb_load(xbase + k0 + 8);
    };
    auto commit_w = [&](const sk4_i32x4* r) {
#pragma unroll
        for (int p = 0; p < 2; ++p)
            *reinterpret_cast<sk4_i32x4*>(wt + st_row * PITCH4 + st_off + p * 8) = r[p];
    };
    auto commit_x = [&](const sk4_i32x4* r) {
#pragma unroll
        for (int p = 0; p < 2; ++p)
            *reinterpret_cast<sk4_i32x4*>(xt + st_row * PITCH4 + st_off + p * 8) = r[p];
    };

    sk4_f32x4 acc[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[n] = sk4_f32x4{0.f, 0.f, 0.f, 0.f};

    const int nchunks = K / KC4;
    sk4_i32x4 w0[2], w1[2], w2[2], w3[2], w4[2], w5[2];
    sk4_i32x4 x0[2], x1[2], x2[2], x3[2], x4[2], x5[2];
    load_w(0, w0); load_w(KC4, w1); load_w(2 * KC4, w2); load_w(3 * KC4, w3);
    load_w(4 * KC4, w4); load_w(5 * KC4, w5);
    load_x(0, x0); load_x(KC4, x1); load_x(2 * KC4, x2);
    load_x(3 * KC4, x3); load_x(4 * KC4, x4); load_x(5 * KC4, x5);
    SK4_WAIT(0);             // one-time drain: makes the uniform steady
    commit_w(w0);            // waits below safe for the early bodies
    commit_x(x0);

    auto iter_body = [&](int i, sk4_i32x4* w_reuse, sk4_i32x4* w_next,
                         sk4_i32x4* x_reuse, sk4_i32x4* x_next) {
        __syncthreads();                 // chunk i visible in LDS
#pragma unroll
        for (int ks = 0; ks < KC4 / 32; ++ks) {
            const vec8 afrag = *reinterpret_cast<const vec8*>(
                xt + (wave * 16 + col) * PITCH4 + ks * 32 + kseg * 8);
#pragma unroll
            for (int n = 0; n < 4; ++n) {
                const vec8 bfrag = *reinterpret_cast<const vec8*>(
                    wt + (n * 16 + col) * PITCH4 + ks * 32 + kseg * 8);
                acc[n] = MF::mfma(afrag, bfrag, acc[n]);
            }
        }
        __syncthreads();                 // LDS reads done before restage
        if (i + 1 < nchunks) {
            SK4_WAIT(18);                // W chunk i+1 landed (see header)
            commit_w(w_next);
            SK4_WAIT(16);                // x chunk i+1 landed
            commit_x(x_next);
        }
        // loads unconditional + clamped: counts stay static; clamped
        // duplicates land in stages that are never committed
        load_w(min(i + 6, nchunks - 1) * KC4, w_reuse);
        load_x(min(i + 6, nchunks - 1) * KC4, x_reuse);
        __builtin_amdgcn_sched_barrier(0);
    };
    for (int i = 0; i < nchunks; i += 6) {
        iter_body(i + 0, w0, w1, x0, x1);
        if (i + 1 < nchunks) iter_body(i + 1, w1, w2, x1, x2);
        if (i + 2 < nchunks) iter_body(i + 2, w2, w3, x2, x3);
        if (i + 3 < nchunks) iter_body(i + 3, w3, w4, x3, x4);
        if (i + 4 < nchunks) iter_body(i + 4, w4, w5, x4, x5);
        if (i + 5 < nchunks) iter_body(i + 5, w5, w0, x5, x0);
    }
    SK4_WAIT(0);             // drain the trailing clamped loads

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
void launch_sk4(torch::Tensor& y, const torch::Tensor& x, const torch::Tensor& w,
                const torch::Tensor* bias) {
    const int M = x.size(0), K = x.size(1), N = w.size(0);
    auto stream = at::hip::getCurrentHIPStream();
    dim3 grid(N / NT4, (M + MT4 - 1) / MT4);
    hipLaunchKernelGGL((skinny_gemm4_kernel<T>), grid, dim3(256), 0, stream,
                       (T*)y.data_ptr(), (const T*)x.data_ptr(),
                       (const T*)w.data_ptr(),
                       bias ? (const T*)bias->data_ptr() : nullptr, M, N, K);
    HIP_CHECK_KERNEL();
}

}  // namespace

void skinny_gemm4(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                  c10::optional<torch::Tensor> bias_opt) {
    const int M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(M <= 1024 && N % NT4 == 0 && K % KC4 == 0 && K >= 6 * KC4,
                "skinny_gemm4: M<=1024, N%64, K%64, K>=384");
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && y.is_contiguous());
    torch::Tensor bias_t;
    const torch::Tensor* bias = nullptr;
    if (bias_opt.has_value()) {
        bias_t = bias_opt->contiguous();
        bias = &bias_t;
    }
    if (x.scalar_type() == at::ScalarType::BFloat16) {
        launch_sk4<__hip_bfloat16>(y, x, w, bias);
    } else if (x.scalar_type() == at::ScalarType::Half) {
        launch_sk4<__half>(y, x, w, bias);
    } else {
        TORCH_CHECK(false, "skinny_gemm4: bf16/f16 only");
    }
}
