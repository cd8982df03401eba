// Fused MoE routing kernel for gfx950 (Mixtral — BASELINE config 5).
//
// Replaces the torch glue on the decode hot path (top-k + softmax + one-hot
// + cumsum-rank + index arithmetic was ~5.6 ms/step at batch 256 in
// profiles/r01 — torch's scan_outer_dim alone was 3 ms on a [T*K, E]
// one-hot): one launch computes, per token row,
//
//   * the top-k experts of the router logits (value order, ties by lower
//     expert id — matches torch.topk),
//   * softmax weights over the selected k logits (fp32),
//   * a padded-buffer destination row dst = expert * C + rank, where rank
//     is drawn from a per-expert atomic counter — dispatch order within an
//     expert is NOT semantically meaningful (the gather back uses the same
//     dst), so no stable scan is needed.
//
// One workgroup; per-expert counters live in LDS.  Decode-sized T only
// (the prefill path uses the sorted per-expert torch route).
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace igw {

template <int BLOCK, int K>
__global__ __launch_bounds__(BLOCK) void moe_route_kernel(
    float* __restrict__ weights,       // [T, K] fp32 softmax weights
    int32_t* __restrict__ dst,         // [T*K] padded-buffer row index
    int32_t* __restrict__ topk_idx,    // [T, K] selected experts
    int32_t* __restrict__ counts,      // [E] per-expert row counts
    const float* __restrict__ logits,  // [T, E] fp32 router logits
    int T, int E, int C) {
  __shared__ uint32_t counters[64];
  for (int e = threadIdx.x; e < 64; e += BLOCK) counters[e] = 0;
  __syncthreads();

  for (int t = threadIdx.x; t < T; t += BLOCK) {
    const float* row = logits + (int64_t)t * E;
    // K is a template constant, so sel/selv stay in registers (runtime
    // indexing of a local array would spill to scratch)
    int sel[K];
    float selv[K];
    float m = -INFINITY;
#pragma unroll
    for (int k = 0; k < K; ++k) {
      int best = 0;
      float bv = -INFINITY;
      for (int e = 0; e < E; ++e) {
        bool taken = false;
#pragma unroll
        for (int k2 = 0; k2 < K; ++k2) taken |= (k2 < k && sel[k2] == e);
        const float v = row[e];
        if (!taken && v > bv) {
          bv = v;
          best = e;
        }
      }
      sel[k] = best;
      selv[k] = bv;
      m = fmaxf(m, bv);
    }
    float z = 0.0f;
    float w[K];
#pragma unroll
    for (int k = 0; k < K; ++k) {
      w[k] = __expf(selv[k] - m);
      z += w[k];
    }
    const float inv = 1.0f / z;
#pragma unroll
    for (int k = 0; k < K; ++k) {
      const int e = sel[k];
      const uint32_t r = atomicAdd(&counters[e], 1u);
      weights[(int64_t)t * K + k] = w[k] * inv;
      topk_idx[(int64_t)t * K + k] = e;
      dst[(int64_t)t * K + k] = e * C + (int32_t)r;
    }
  }
  __syncthreads();
  for (int e = threadIdx.x; e < E; e += BLOCK)
    counts[e] = (int32_t)counters[e];
}

void moe_route(torch::Tensor weights, torch::Tensor dst,
               torch::Tensor topk_idx, torch::Tensor counts,
               torch::Tensor logits, int64_t C) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kFloat32);
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(weights.dtype() == torch::kFloat32);
  TORCH_CHECK(dst.dtype() == torch::kInt32 &&
              topk_idx.dtype() == torch::kInt32 &&
              counts.dtype() == torch::kInt32);
  const int T = logits.size(0);
  const int E = logits.size(1);
  const int K = weights.size(1);
  TORCH_CHECK(E <= 64, "moe_route supports up to 64 experts");
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  constexpr int BLOCK = 256;
#define LAUNCH_ROUTE(KV)                                                     \
  hipLaunchKernelGGL((moe_route_kernel<BLOCK, KV>), dim3(1), dim3(BLOCK), 0, \
                     stream, (float*)weights.data_ptr(),                     \
                     (int32_t*)dst.data_ptr(), (int32_t*)topk_idx.data_ptr(),\
                     (int32_t*)counts.data_ptr(),                            \
                     (const float*)logits.data_ptr(), T, E, (int)C)
  switch (K) {
    case 1: LAUNCH_ROUTE(1); break;
    case 2: LAUNCH_ROUTE(2); break;
    case 4: LAUNCH_ROUTE(4); break;
    case 8: LAUNCH_ROUTE(8); break;
    default: TORCH_CHECK(false, "unsupported top-k for moe_route: ", K);
  }
#undef LAUNCH_ROUTE
  HIP_CHECK_KERNEL();
}

}  // namespace igw
