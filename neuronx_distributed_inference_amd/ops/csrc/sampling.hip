// Fused top-k / top-p / temperature / inverse-CDF sampling (K8/K9/K17-analog,
// ref modules/generation/sampling.py:285-370).
//
// Input is the SORTED-descending candidate matrix (B, K<=256) produced by the
// distributed top-k stage; one wave per row does temperature scaling,
// softmax, the top-p cutoff on the exclusive prefix sum, and the inverse-CDF
// pick in registers (4 candidates per lane). Returns the candidate INDEX
// (position in the sorted list); the caller maps back to vocab ids.

#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

template <int KMAX>
__global__ void topk_topp_sample_kernel(
    long* __restrict__ out,            // (B,)
    const float* __restrict__ vals,    // (B, K) sorted desc
    const int* __restrict__ top_k,     // (B,)
    const float* __restrict__ top_p,   // (B,)
    const float* __restrict__ temp,    // (B,)
    const float* __restrict__ rand,    // (B,)
    const int K) {
    constexpr int PER_LANE = KMAX / WAVE_SIZE;
    const int b = blockIdx.x;
    const int lane = threadIdx.x;

    int kk = top_k[b];
    if (kk == 1) {  // greedy: first (max) candidate
        if (lane == 0) out[b] = 0;
        return;
    }
    if (kk <= 0 || kk > K) kk = K;
    const float invt = 1.f / fmaxf(temp[b], 1e-6f);
    const float p_thresh = top_p[b];

    // each lane holds candidates [lane*PER_LANE, ...) — contiguous per lane
    float v[PER_LANE];
    const float vmax = vals[(long)b * K + 0];  // sorted desc
#pragma unroll
    for (int j = 0; j < PER_LANE; ++j) {
        const int idx = lane * PER_LANE + j;
        v[j] = (idx < kk) ? __expf((vals[(long)b * K + idx] - vmax) * invt) : 0.f;
    }
    // local inclusive prefix + wave-exclusive scan of lane sums
    float lane_sum = 0.f;
#pragma unroll
    for (int j = 0; j < PER_LANE; ++j) lane_sum += v[j];
    float excl = 0.f;
    {
        float run = lane_sum;
#pragma unroll
        for (int off = 1; off < WAVE_SIZE; off <<= 1) {
            const float o = __shfl_up(run, off, WAVE_SIZE);
            if (lane >= off) run += o;
        }
        excl = run - lane_sum;  // exclusive prefix of lane sums
    }
    const float total = wave_reduce_sum(lane_sum);

    // top-p mask: keep candidate i while its EXCLUSIVE cdf over the original
    // distribution is < p * total (the first candidate is always kept) —
    // exactly the reference's (csum - probs) < p rule.
    float pref = excl;
#pragma unroll
    for (int j = 0; j < PER_LANE; ++j) {
        const float orig = v[j];
        const bool keep = (pref < p_thresh * total) && (orig > 0.f);
        pref += orig;
        if (!keep) v[j] = 0.f;
    }
    // rebuild the prefix over the kept mass
    float lane_sum2 = 0.f;
#pragma unroll
    for (int j = 0; j < PER_LANE; ++j) lane_sum2 += v[j];
    float excl2;
    {
        float run = lane_sum2;
#pragma unroll
        for (int off = 1; off < WAVE_SIZE; off <<= 1) {
            const float o = __shfl_up(run, off, WAVE_SIZE);
            if (lane >= off) run += o;
        }
        excl2 = run - lane_sum2;
    }
    const float total2 = wave_reduce_sum(lane_sum2);
    const float target = rand[b] * total2;

    // pick the first candidate whose inclusive cdf >= target
    int pick = KMAX;
    float cdf = excl2;
#pragma unroll
    for (int j = 0; j < PER_LANE; ++j) {
        cdf += v[j];
        if (v[j] > 0.f && cdf >= target && pick == KMAX) pick = lane * PER_LANE + j;
    }
    // wave-min of pick
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        pick = min(pick, __shfl_xor(pick, off, WAVE_SIZE));
    if (lane == 0) out[b] = (pick >= KMAX) ? (kk - 1) : pick;
}

void topk_topp_sample(torch::Tensor out, torch::Tensor vals, torch::Tensor top_k,
                      torch::Tensor top_p, torch::Tensor temp, torch::Tensor rand,
                      long global_topk) {
    const int B = vals.size(0);
    const int K = vals.size(1);
    TORCH_CHECK(K <= 256, "candidate count must be <= 256, got ", K);
    TORCH_CHECK(vals.scalar_type() == at::ScalarType::Float, "vals must be fp32");
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL((topk_topp_sample_kernel<256>), dim3(B), dim3(WAVE_SIZE), 0, stream,
                       out.data_ptr<long>(), vals.data_ptr<float>(),
                       top_k.data_ptr<int>(), top_p.data_ptr<float>(),
                       temp.data_ptr<float>(), rand.data_ptr<float>(), K);
    HIP_CHECK_KERNEL();
}
