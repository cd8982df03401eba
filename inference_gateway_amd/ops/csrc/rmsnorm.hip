// RMSNorm kernels for gfx950.
//
// Reference behavior parity: this is part of the net-new local inference
// engine (SURVEY.md §0/§7 — the Go reference has no kernels); numerics follow
// the standard Llama RMSNorm: y = x / sqrt(mean(x^2) + eps) * w, with the
// mean computed in fp32.
//
// MI355X design: memory-bound streaming op — one workgroup (256 threads) per
// row, bf16x8 (16-byte) vector loads/stores (scalar bf16 loads halve HBM
// throughput on this chip), fp32 accumulation, block reduce through LDS.
// The fused variant folds the residual add into the same pass so the row is
// read once and written once (residual updated in place).

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace igw {

template <int BLOCK>
__global__ void rmsnorm_kernel(uint16_t* __restrict__ out,
                               const uint16_t* __restrict__ in,
                               const uint16_t* __restrict__ weight,
                               int hidden, float eps) {
  __shared__ float red[BLOCK / WAVE_SIZE];
  const int row = blockIdx.x;
  const uint16_t* x = in + (int64_t)row * hidden;
  uint16_t* y = out + (int64_t)row * hidden;

  float ss = 0.0f;
  // 8 bf16 per thread per iteration (16B vector loads)
  for (int i = threadIdx.x * 8; i < hidden; i += BLOCK * 8) {
    ushort8 v = *reinterpret_cast<const ushort8*>(x + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(v.v[j]);
      ss += f * f;
    }
  }
  ss = block_reduce_sum<BLOCK>(ss, red);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x * 8; i < hidden; i += BLOCK * 8) {
    ushort8 v = *reinterpret_cast<const ushort8*>(x + i);
    ushort8 w = *reinterpret_cast<const ushort8*>(weight + i);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      o.v[j] = f32_to_bf16(bf16_to_f32(v.v[j]) * inv * bf16_to_f32(w.v[j]));
    }
    *reinterpret_cast<ushort8*>(y + i) = o;
  }
}

// x <- rmsnorm(x + residual) * w ; residual <- x + residual
//
// N_ITER variant: the whole row stays in registers across the block
// reduce (x, residual and weight are each read ONCE from HBM; no re-read
// after the barrier), with all loads issued up front for memory-level
// parallelism — this op runs at 2 workgroups/CU so latency, not
// bandwidth, is the limit.
template <int BLOCK, int N_ITER>
__global__ void fused_add_rmsnorm_kernel(uint16_t* __restrict__ x,
                                         uint16_t* __restrict__ residual,
                                         const uint16_t* __restrict__ weight,
                                         int hidden, float eps) {
  __shared__ float red[BLOCK / WAVE_SIZE];
  const int row = blockIdx.x;
  uint16_t* xr = x + (int64_t)row * hidden;
  uint16_t* rr = residual + (int64_t)row * hidden;

  ushort8 a[N_ITER], b[N_ITER], w[N_ITER];
#pragma unroll
  for (int it = 0; it < N_ITER; ++it) {
    const int i = threadIdx.x * 8 + it * BLOCK * 8;
    a[it] = *reinterpret_cast<const ushort8*>(xr + i);
    b[it] = *reinterpret_cast<const ushort8*>(rr + i);
    w[it] = *reinterpret_cast<const ushort8*>(weight + i);
  }
  float ss = 0.0f;
  ushort8 s[N_ITER];
#pragma unroll
  for (int it = 0; it < N_ITER; ++it) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(a[it].v[j]) + bf16_to_f32(b[it].v[j]);
      s[it].v[j] = f32_to_bf16(f);
      // accumulate on the bf16-rounded sum so the normalizer matches the
      // stored residual exactly (same as the torch reference path)
      float fr = bf16_to_f32(s[it].v[j]);
      ss += fr * fr;
    }
    *reinterpret_cast<ushort8*>(rr + threadIdx.x * 8 + it * BLOCK * 8) =
        s[it];
  }
  ss = block_reduce_sum<BLOCK>(ss, red);
  const float inv = rsqrtf(ss / hidden + eps);

#pragma unroll
  for (int it = 0; it < N_ITER; ++it) {
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      o.v[j] =
          f32_to_bf16(bf16_to_f32(s[it].v[j]) * inv * bf16_to_f32(w[it].v[j]));
    }
    *reinterpret_cast<ushort8*>(xr + threadIdx.x * 8 + it * BLOCK * 8) = o;
  }
}

// generic fallback for hidden sizes that are not 1/2/4 x BLOCK*8
template <int BLOCK>
__global__ void fused_add_rmsnorm_generic_kernel(
    uint16_t* __restrict__ x, uint16_t* __restrict__ residual,
    const uint16_t* __restrict__ weight, int hidden, float eps) {
  __shared__ float red[BLOCK / WAVE_SIZE];
  const int row = blockIdx.x;
  uint16_t* xr = x + (int64_t)row * hidden;
  uint16_t* rr = residual + (int64_t)row * hidden;

  float ss = 0.0f;
  for (int i = threadIdx.x * 8; i < hidden; i += BLOCK * 8) {
    ushort8 a = *reinterpret_cast<const ushort8*>(xr + i);
    ushort8 b = *reinterpret_cast<const ushort8*>(rr + i);
    ushort8 s;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(a.v[j]) + bf16_to_f32(b.v[j]);
      s.v[j] = f32_to_bf16(f);
      float fr = bf16_to_f32(s.v[j]);
      ss += fr * fr;
    }
    *reinterpret_cast<ushort8*>(rr + i) = s;
  }
  ss = block_reduce_sum<BLOCK>(ss, red);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x * 8; i < hidden; i += BLOCK * 8) {
    ushort8 s = *reinterpret_cast<const ushort8*>(rr + i);
    ushort8 w = *reinterpret_cast<const ushort8*>(weight + i);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      o.v[j] = f32_to_bf16(bf16_to_f32(s.v[j]) * inv * bf16_to_f32(w.v[j]));
    }
    *reinterpret_cast<ushort8*>(xr + i) = o;
  }
}

void rmsnorm(torch::Tensor out, torch::Tensor in, torch::Tensor weight,
             double eps) {
  TORCH_CHECK(in.is_cuda() && in.dtype() == torch::kBFloat16);
  TORCH_CHECK(in.is_contiguous() && out.is_contiguous());
  const int hidden = in.size(-1);
  const int rows = in.numel() / hidden;
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  constexpr int BLOCK = 256;
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((rmsnorm_kernel<BLOCK>), dim3(rows), dim3(BLOCK), 0,
                     stream, (uint16_t*)out.data_ptr(),
                     (const uint16_t*)in.data_ptr(),
                     (const uint16_t*)weight.data_ptr(), hidden, (float)eps);
  HIP_CHECK_KERNEL();
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor weight, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && residual.is_contiguous());
  const int hidden = x.size(-1);
  const int rows = x.numel() / hidden;
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  constexpr int BLOCK = 256;
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  const int per = BLOCK * 8;
  if (hidden == per) {
    hipLaunchKernelGGL((fused_add_rmsnorm_kernel<BLOCK, 1>), dim3(rows),
                       dim3(BLOCK), 0, stream, (uint16_t*)x.data_ptr(),
                       (uint16_t*)residual.data_ptr(),
                       (const uint16_t*)weight.data_ptr(), hidden, (float)eps);
  } else if (hidden == 2 * per) {
    hipLaunchKernelGGL((fused_add_rmsnorm_kernel<BLOCK, 2>), dim3(rows),
                       dim3(BLOCK), 0, stream, (uint16_t*)x.data_ptr(),
                       (uint16_t*)residual.data_ptr(),
                       (const uint16_t*)weight.data_ptr(), hidden, (float)eps);
  } else if (hidden == 4 * per) {
    hipLaunchKernelGGL((fused_add_rmsnorm_kernel<BLOCK, 4>), dim3(rows),
                       dim3(BLOCK), 0, stream, (uint16_t*)x.data_ptr(),
                       (uint16_t*)residual.data_ptr(),
                       (const uint16_t*)weight.data_ptr(), hidden, (float)eps);
  } else {
    hipLaunchKernelGGL((fused_add_rmsnorm_generic_kernel<BLOCK>), dim3(rows),
                       dim3(BLOCK), 0, stream, (uint16_t*)x.data_ptr(),
                       (uint16_t*)residual.data_ptr(),
                       (const uint16_t*)weight.data_ptr(), hidden, (float)eps);
  }
  HIP_CHECK_KERNEL();
}

}  // namespace igw
