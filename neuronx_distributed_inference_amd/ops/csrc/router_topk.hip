// Fused MoE router selection (SURVEY §2.7 router analog): ONE kernel
// replaces the torch topk chain (gatherTopK + bitonicSort + softmax +
// div — measured ~0.9 ms/step of launch latency on Mixtral b256 decode,
// rocprofv3 r02_moe_quant_kernel_stats.md). One thread per token row:
// register insertion-select of the top-k (k <= 8) over E logits, then
// the router activation (post-softmax over k | full-E pre-softmax |
// sigmoid) and optional sum-normalization — all in fp32, matching
// modules/moe.py MoERouter semantics.

#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int KMAX = 8;

// ONE WAVE per token row (the first cut ran one THREAD per row and
// serialized the E-loop — 193 us/launch at E=128, 24% of the Qwen3-MoE
// step). Selection = k rounds of wave-argmax: each lane scans its
// strided slice of the E logits (excluding already-chosen indices) and
// a 6-step shuffle reduction finds the global (value, lowest-index)
// max, broadcast to all lanes.
__global__ __launch_bounds__(256) void router_topk_kernel(
    float* __restrict__ vals,        // (T, k)
    long* __restrict__ idx,          // (T, k)
    const float* __restrict__ logits,  // (T, E)
    const int T, const int E, const int k,
    const int act_sigmoid, const int pre_softmax, const int normalize) {
    const int wid = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int row = blockIdx.x * 4 + wid;
    if (row >= T) return;
    const float* lr = logits + (long)row * E;

    float cv[KMAX];
    int ci[KMAX];
#pragma unroll
    for (int j = 0; j < KMAX; ++j) {
        cv[j] = -1e30f;
        ci[j] = -1;
    }
    for (int round = 0; round < k; ++round) {
        float lv = -1e30f;
        int li = E;  // sentinel larger than any index
        for (int e = lane; e < E; e += 64) {
            bool taken = false;
#pragma unroll
            for (int j = 0; j < KMAX; ++j)
                if (j < round) taken = taken || (e == ci[j]);
            const float v = lr[e];
            if (!taken && (v > lv || (v == lv && e < li))) {
                lv = v;
                li = e;
            }
        }
#pragma unroll
        for (int off = 32; off >= 1; off >>= 1) {
            const float ov = __shfl_down(lv, off, 64);
            const int oi = __shfl_down(li, off, 64);
            if (ov > lv || (ov == lv && oi < li)) {
                lv = ov;
                li = oi;
            }
        }
        lv = __shfl(lv, 0, 64);
        li = __shfl(li, 0, 64);
        cv[round] = lv;
        ci[round] = li;
    }

    if (pre_softmax) {
        // full-E softmax: cv[0] IS the max — snapshot before the
        // in-place exp overwrites it
        const float fm = cv[0];
        float ls = 0.f;
        for (int e = lane; e < E; e += 64) ls += __expf(lr[e] - fm);
#pragma unroll
        for (int off = 32; off >= 1; off >>= 1)
            ls += __shfl_down(ls, off, 64);
        ls = __shfl(ls, 0, 64);
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) cv[j] = __expf(cv[j] - fm) / ls;
    } else if (act_sigmoid) {
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) cv[j] = 1.f / (1.f + __expf(-cv[j]));
    } else {
        const float m0 = cv[0];
        float s = 0.f;
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) {
                cv[j] = __expf(cv[j] - m0);
                s += cv[j];
            }
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) cv[j] /= s;
    }
    if (normalize) {
        float s = 0.f;
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) s += cv[j];
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) cv[j] /= fmaxf(s, 1e-20f);
    }
    if (lane == 0) {
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) {
                vals[(long)row * k + j] = cv[j];
                idx[(long)row * k + j] = ci[j];
            }
    }
}

}  // namespace

void router_topk(torch::Tensor vals, torch::Tensor idx, torch::Tensor logits,
                 long top_k, long act_sigmoid, long pre_softmax,
                 long normalize) {
    TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous() &&
                logits.scalar_type() == at::ScalarType::Float, "logits f32");
    const int T = logits.size(0), E = logits.size(1);
    TORCH_CHECK(top_k >= 1 && top_k <= KMAX, "top_k <= 8");
    TORCH_CHECK(vals.numel() == T * top_k && idx.numel() == T * top_k &&
                vals.scalar_type() == at::ScalarType::Float &&
                idx.scalar_type() == at::ScalarType::Long, "out shapes");
    auto stream = at::hip::getCurrentHIPStream();
    const int threads = 256;          // 4 waves; one wave per row
    const int blocks = (T + 3) / 4;
    hipLaunchKernelGGL(router_topk_kernel, dim3(blocks), dim3(threads), 0,
                       stream, vals.data_ptr<float>(), idx.data_ptr<long>(),
                       logits.data_ptr<float>(), T, E, (int)top_k,
                       (int)act_sigmoid, (int)pre_softmax, (int)normalize);
    HIP_CHECK_KERNEL();
}
