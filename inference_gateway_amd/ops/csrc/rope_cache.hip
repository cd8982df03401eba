// Rotary position embedding + paged-KV-cache append kernels for gfx950.
//
// Part of the net-new local engine (SURVEY.md §7 stage 3b). Both ops are
// memory-bound elementwise passes, fused as tightly as the call graph allows:
// RoPE rotates Q and K in one launch; the cache append scatters K and V of
// the current step into the paged cache ([num_blocks, n_kv_heads, block_size,
// head_dim] layout chosen so the decode-attention kernel reads contiguous
// [block_size, head_dim] tiles per (block, head)).

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace igw {

// GPT-NeoX-style rotate-half RoPE, in place on q and k.
// q: [T, NQ*HD], k: [T, NKV*HD], cos_sin: [max_pos, HD] fp32
// (cos in [0, HD/2), sin in [HD/2, HD)).
__global__ void rope_kernel(uint16_t* __restrict__ q, uint16_t* __restrict__ k,
                            const int32_t* __restrict__ positions,
                            const float* __restrict__ cos_sin, int nq, int nkv,
                            int hd, int qs, int ks) {
  // qs/ks: row strides in elements — q/k may be views into a fused-QKV
  // GEMM output (no .contiguous() copies on the decode hot path).
  // Vectorized 8 pairs per thread: the rotate-half halves are each
  // contiguous, so x1/x2 load as ushort8 and cos/sin as two float4s
  // (scalar 2-B pair loads ran at 30% of the HBM roofline).
  const int t = blockIdx.x;
  const int pos = positions[t];
  const float* cs = cos_sin + (int64_t)pos * hd;
  const int half = hd / 2;
  const int nheads = nq + nkv;
  const int groups = half >> 3;       // 8 pairs per group (hd % 16 == 0)
  for (int i = threadIdx.x; i < nheads * groups; i += blockDim.x) {
    const int h = i / groups;
    const int d0 = (i % groups) * 8;
    uint16_t* base = (h < nq) ? q + (int64_t)t * qs + h * hd
                              : k + (int64_t)t * ks + (h - nq) * hd;
    const ushort8 v1 = *reinterpret_cast<const ushort8*>(base + d0);
    const ushort8 v2 = *reinterpret_cast<const ushort8*>(base + half + d0);
    const float4 c0 = *reinterpret_cast<const float4*>(cs + d0);
    const float4 c1 = *reinterpret_cast<const float4*>(cs + d0 + 4);
    const float4 s0 = *reinterpret_cast<const float4*>(cs + half + d0);
    const float4 s1 = *reinterpret_cast<const float4*>(cs + half + d0 + 4);
    const float cv[8] = {c0.x, c0.y, c0.z, c0.w, c1.x, c1.y, c1.z, c1.w};
    const float sv[8] = {s0.x, s0.y, s0.z, s0.w, s1.x, s1.y, s1.z, s1.w};
    ushort8 o1, o2;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x1 = bf16_to_f32(v1.v[j]);
      const float x2 = bf16_to_f32(v2.v[j]);
      o1.v[j] = f32_to_bf16(x1 * cv[j] - x2 * sv[j]);
      o2.v[j] = f32_to_bf16(x2 * cv[j] + x1 * sv[j]);
    }
    *reinterpret_cast<ushort8*>(base + d0) = o1;
    *reinterpret_cast<ushort8*>(base + half + d0) = o2;
  }
}

// Scatter this step's K/V into the paged cache.
// k/v: [T, NKV, HD]; caches: [NB, NKV, BS, HD]; slot_mapping: [T] int32
// (slot = block_id * BS + offset; slot < 0 => skip, padding token).
__global__ void append_kv_kernel(const uint16_t* __restrict__ k,
                                 const uint16_t* __restrict__ v,
                                 uint16_t* __restrict__ kcache,
                                 uint16_t* __restrict__ vcache,
                                 const int32_t* __restrict__ slot_mapping,
                                 int nkv, int hd, int block_size, int ks,
                                 int vs) {
  const int t = blockIdx.x;
  const int slot = slot_mapping[t];
  if (slot < 0) return;
  const int64_t blk = slot / block_size;
  const int off = slot % block_size;
  // dest tile base for (blk, head, off): ((blk*NKV + h)*BS + off)*HD
  for (int i = threadIdx.x * 8; i < nkv * hd; i += blockDim.x * 8) {
    const int h = i / hd;
    const int d = i % hd;
    const int64_t dst = (((int64_t)blk * nkv + h) * block_size + off) * hd + d;
    *reinterpret_cast<ushort8*>(kcache + dst) =
        *reinterpret_cast<const ushort8*>(k + (int64_t)t * ks + h * hd + d);
    *reinterpret_cast<ushort8*>(vcache + dst) =
        *reinterpret_cast<const ushort8*>(v + (int64_t)t * vs + h * hd + d);
  }
}

void rotary_embedding(torch::Tensor positions, torch::Tensor q, torch::Tensor k,
                      int64_t head_dim, torch::Tensor cos_sin_cache) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(positions.dtype() == torch::kInt32);
  TORCH_CHECK(cos_sin_cache.dtype() == torch::kFloat32);
  TORCH_CHECK(q.stride(-1) == 1 && k.stride(-1) == 1);
  TORCH_CHECK(head_dim % 16 == 0, "rope: head_dim must be 16-aligned");
  const int T = q.size(0);
  const int hd = head_dim;
  const int nq = q.size(-1) / hd;
  const int nkv = k.size(-1) / hd;
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  const int threads = 256;
  hipLaunchKernelGGL(rope_kernel, dim3(T), dim3(threads), 0, stream,
                     (uint16_t*)q.data_ptr(), (uint16_t*)k.data_ptr(),
                     (const int32_t*)positions.data_ptr(),
                     (const float*)cos_sin_cache.data_ptr(), nq, nkv, hd,
                     (int)q.stride(0), (int)k.stride(0));
  HIP_CHECK_KERNEL();
}

void append_kv_cache(torch::Tensor k, torch::Tensor v, torch::Tensor kcache,
                     torch::Tensor vcache, torch::Tensor slot_mapping) {
  TORCH_CHECK(k.is_cuda() && k.dtype() == torch::kBFloat16);
  TORCH_CHECK(slot_mapping.dtype() == torch::kInt32);
  // k/v: [T, NKV, HD] views with possibly-strided row dim (fused QKV out)
  TORCH_CHECK(k.stride(-1) == 1 && v.stride(-1) == 1);
  TORCH_CHECK(k.stride(1) == k.size(2) && v.stride(1) == v.size(2));
  TORCH_CHECK(kcache.is_contiguous() && vcache.is_contiguous());
  const int T = k.size(0);
  const int nkv = kcache.size(1);
  const int block_size = kcache.size(2);
  const int hd = kcache.size(3);
  TORCH_CHECK(hd % 8 == 0);
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(append_kv_kernel, dim3(T), dim3(256), 0, stream,
                     (const uint16_t*)k.data_ptr(),
                     (const uint16_t*)v.data_ptr(), (uint16_t*)kcache.data_ptr(),
                     (uint16_t*)vcache.data_ptr(),
                     (const int32_t*)slot_mapping.data_ptr(), nkv, hd,
                     block_size, (int)k.stride(0), (int)v.stride(0));
  HIP_CHECK_KERNEL();
}

}  // namespace igw
