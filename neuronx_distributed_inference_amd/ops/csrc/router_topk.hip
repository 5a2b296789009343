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

__global__ void router_topk_kernel(
    float* __restrict__ vals,        // (T, k)
    long* __restrict__ idx,          // (T, k)
    const float* __restrict__ logits,  // (T, E)
    const int T, const int E, const int k,
    const int act_sigmoid, const int pre_softmax, const int normalize) {
    const int row = blockIdx.x * blockDim.x + threadIdx.x;
    if (row >= T) return;
    const float* lr = logits + (long)row * E;

    float bv[KMAX];
    int bi[KMAX];
#pragma unroll
    for (int j = 0; j < KMAX; ++j) {
        bv[j] = -1e30f;
        bi[j] = -1;
    }
    float fmax = -1e30f, fsum = 0.f;
    for (int e = 0; e < E; ++e) {
        const float v = lr[e];
        fmax = fmaxf(fmax, v);
        // descending insertion (stable: strict > keeps the earlier index
        // on ties, matching torch.topk's lower-index preference)
#pragma unroll
        for (int j = 0; j < KMAX; ++j) {
            if (j >= k) break;
            if (v > bv[j]) {
                for (int m = k - 1; m > j; --m) {
                    bv[m] = bv[m - 1];
                    bi[m] = bi[m - 1];
                }
                bv[j] = v;
                bi[j] = e;
                break;
            }
        }
    }
    if (pre_softmax) {
        for (int e = 0; e < E; ++e) fsum += __expf(lr[e] - fmax);
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) bv[j] = __expf(bv[j] - fmax) / fsum;
    } else if (act_sigmoid) {
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) bv[j] = 1.f / (1.f + __expf(-bv[j]));
    } else {
        // softmax over the selected k (bv[0] is the max — snapshot it
        // BEFORE the in-place exp overwrites it)
        const float m0 = bv[0];
        float s = 0.f;
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) {
                bv[j] = __expf(bv[j] - m0);
                s += bv[j];
            }
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) bv[j] /= s;
    }
    if (normalize) {
        float s = 0.f;
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) s += bv[j];
#pragma unroll
        for (int j = 0; j < KMAX; ++j)
            if (j < k) bv[j] /= fmaxf(s, 1e-20f);
    }
#pragma unroll
    for (int j = 0; j < KMAX; ++j)
        if (j < k) {
            vals[(long)row * k + j] = bv[j];
            idx[(long)row * k + j] = bi[j];
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
    const int threads = 256;
    const int blocks = (T + threads - 1) / threads;
    hipLaunchKernelGGL(router_topk_kernel, dim3(blocks), dim3(threads), 0,
                       stream, vals.data_ptr<float>(), idx.data_ptr<long>(),
                       logits.data_ptr<float>(), T, E, (int)top_k,
                       (int)act_sigmoid, (int)pre_softmax, (int)normalize);
    HIP_CHECK_KERNEL();
}
