// Fused SwiGLU activation for gfx950: out = silu(gate) * up.
//
// Memory-bound elementwise op; the gate/up halves come straight out of the
// fused gate_up projection GEMM ([T, 2*I] with gate in [0, I)), so this is
// one read + one write of the intermediate activation instead of three
// separate passes. bf16x8 vector loads, fp32 math.

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace igw {

__global__ void silu_mul_kernel(uint16_t* __restrict__ out,
                                const uint16_t* __restrict__ gate_up,
                                int inter) {
  const int row = blockIdx.x;
  const uint16_t* g = gate_up + (int64_t)row * 2 * inter;
  const uint16_t* u = g + inter;
  uint16_t* y = out + (int64_t)row * inter;
  for (int i = threadIdx.x * 8; i < inter; i += blockDim.x * 8) {
    ushort8 gv = *reinterpret_cast<const ushort8*>(g + i);
    ushort8 uv = *reinterpret_cast<const ushort8*>(u + i);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf16_to_f32(gv.v[j]);
      const float s = x / (1.0f + __expf(-x));
      o.v[j] = f32_to_bf16(s * bf16_to_f32(uv.v[j]));
    }
    *reinterpret_cast<ushort8*>(y + i) = o;
  }
}

// Grouped variant for the MoE padded dispatch buffer: rows beyond each
// expert's count are skipped entirely (they were never written by the
// grouped GEMM, and processing them would read uninitialized memory AND
// burn bandwidth on up-to-4x padding).
__global__ void silu_mul_grouped_kernel(uint16_t* __restrict__ out,
                                        const uint16_t* __restrict__ gate_up,
                                        const int32_t* __restrict__ counts,
                                        int capacity, int inter) {
  const int row = blockIdx.x;
  if (row % capacity >= counts[row / capacity]) return;
  const uint16_t* g = gate_up + (int64_t)row * 2 * inter;
  const uint16_t* u = g + inter;
  uint16_t* y = out + (int64_t)row * inter;
  for (int i = threadIdx.x * 8; i < inter; i += blockDim.x * 8) {
    ushort8 gv = *reinterpret_cast<const ushort8*>(g + i);
    ushort8 uv = *reinterpret_cast<const ushort8*>(u + i);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf16_to_f32(gv.v[j]);
      const float s = x / (1.0f + __expf(-x));
      o.v[j] = f32_to_bf16(s * bf16_to_f32(uv.v[j]));
    }
    *reinterpret_cast<ushort8*>(y + i) = o;
  }
}

void silu_mul_grouped(torch::Tensor out, torch::Tensor gate_up,
                      torch::Tensor counts, int64_t capacity) {
  TORCH_CHECK(gate_up.is_cuda() && gate_up.dtype() == torch::kBFloat16);
  TORCH_CHECK(gate_up.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(counts.dtype() == torch::kInt32);
  const int inter = out.size(-1);
  TORCH_CHECK(gate_up.size(-1) == 2 * inter);
  TORCH_CHECK(inter % 8 == 0);
  const int rows = out.numel() / inter;
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(silu_mul_grouped_kernel, dim3(rows), dim3(256), 0,
                     stream, (uint16_t*)out.data_ptr(),
                     (const uint16_t*)gate_up.data_ptr(),
                     (const int32_t*)counts.data_ptr(), (int)capacity,
                     inter);
  HIP_CHECK_KERNEL();
}

void silu_mul(torch::Tensor out, torch::Tensor gate_up) {
  TORCH_CHECK(gate_up.is_cuda() && gate_up.dtype() == torch::kBFloat16);
  TORCH_CHECK(gate_up.is_contiguous() && out.is_contiguous());
  const int inter = out.size(-1);
  TORCH_CHECK(gate_up.size(-1) == 2 * inter);
  TORCH_CHECK(inter % 8 == 0);
  const int rows = out.numel() / inter;
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(silu_mul_kernel, dim3(rows), dim3(256), 0, stream,
                     (uint16_t*)out.data_ptr(),
                     (const uint16_t*)gate_up.data_ptr(), inter);
  HIP_CHECK_KERNEL();
}

}  // namespace igw
