// Paged-attention decode kernel for gfx950 (MI355X).
//
// Net-new engine scope (SURVEY.md §7 stage 3b): single-query attention over a
// paged KV cache, GQA-aware. One workgroup per (sequence, kv_head); the GQA
// group's G query heads ride along so each K/V byte is read exactly once per
// step — decode attention on this chip is HBM-bound, so the kernel is built
// around minimal, fully-coalesced traffic:
//
//   * cache layout [num_blocks, n_kv_heads, block_size=16, head_dim] gives a
//     contiguous 16x128 bf16 tile (4 KiB) per (block, head);
//   * K pass: the wave reads a whole tile with 64 lanes x 64 B (4 lanes per
//     token x 32-dim segment), dot-products against Q (pre-scaled, staged in
//     LDS as fp32), 4-lane shfl reduce -> per-token scores;
//   * online softmax per wave (m, l, acc in registers; fp32);
//   * V pass: lanes switch roles to 2 output dims each; per token the wave
//     reads one contiguous 256 B row of V; probabilities broadcast via LDS;
//   * waves stripe over cache blocks and merge via LDS at the end
//     (flash-decoding style rescale).
//
// Wavefront-64 throughout; no CUDA-isms, no Triton.

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace igw {

template <int HD, int G, int NWAVES>
__global__ __launch_bounds__(NWAVES* WAVE_SIZE) void paged_attn_decode_kernel(
    uint16_t* __restrict__ out,          // [B, NQ, HD]
    const uint16_t* __restrict__ q,      // [B, NQ, HD]
    const uint16_t* __restrict__ kcache, // [NB, NKV, BS, HD]
    const uint16_t* __restrict__ vcache, // [NB, NKV, BS, HD]
    const int32_t* __restrict__ block_table, // [B, max_blocks]
    const int32_t* __restrict__ ctx_lens,    // [B]
    int max_blocks, int nkv, float scale) {
  constexpr int BS = 16;  // tokens per cache block
  const int kvh = blockIdx.x;
  const int seq = blockIdx.y;
  const int nq = nkv * G;
  const int ctx = ctx_lens[seq];
  const int nblocks = (ctx + BS - 1) / BS;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE_SIZE - 1);
  const int wave = tid >> 6;

  // LDS: Q (fp32, pre-scaled), per-wave P broadcast, merge scratch.
  __shared__ float q_lds[G][HD];
  __shared__ float p_lds[NWAVES][G][BS];
  __shared__ float m_lds[NWAVES][G];
  __shared__ float l_lds[NWAVES][G];
  __shared__ float acc_lds[NWAVES][G][HD];

  // stage Q*scale into LDS
  for (int i = tid; i < G * HD; i += NWAVES * WAVE_SIZE) {
    const int j = i / HD;
    const int d = i % HD;
    q_lds[j][d] =
        bf16_to_f32(q[((int64_t)seq * nq + kvh * G + j) * HD + d]) * scale;
  }
  __syncthreads();

  const int tok = lane >> 2;        // 16 tokens per block tile
  const int seg = (lane & 3) * 32;  // 32-dim segment for the K dot
  // V pass: 16 lanes x 8 contiguous dims (one b128 read per lane), 4 token
  // subgroups; per instruction the wave reads 4 complete 256 B V rows.
  const int vlane = lane & 15;      // dim group: dims [vlane*8, vlane*8+8)
  const int vtok = lane >> 4;       // token subgroup: t = vtok + 4*t4

  float m[G], l[G], acc[G][8];
#pragma unroll
  for (int j = 0; j < G; ++j) {
    m[j] = -INFINITY;
    l[j] = 0.0f;
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[j][e] = 0.0f;
  }

  for (int b = wave; b < nblocks; b += NWAVES) {
    const int64_t pb = block_table[(int64_t)seq * max_blocks + b];
    const uint16_t* kbase =
        kcache + (((int64_t)pb * nkv + kvh) * BS) * HD;
    const uint16_t* vbase =
        vcache + (((int64_t)pb * nkv + kvh) * BS) * HD;

    // ---- K pass: 64 B of K per lane (token tok, dims [seg, seg+32)) ----
    float kf[32];
    {
      const uint16_t* kp = kbase + tok * HD + seg;
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        ushort8 kv8 = *reinterpret_cast<const ushort8*>(kp + c * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e) kf[c * 8 + e] = bf16_to_f32(kv8.v[e]);
      }
    }
    const bool valid = (b * BS + tok) < ctx;

    float p[G];
#pragma unroll
    for (int j = 0; j < G; ++j) {
      float s = 0.0f;
#pragma unroll
      for (int d = 0; d < 32; ++d) s += q_lds[j][seg + d] * kf[d];
      s = group_reduce_sum<4>(s);  // full dot at all 4 lanes of the token
      if (!valid) s = -INFINITY;

      // online softmax (per wave)
      const float pmax = wave_reduce_max(s);
      if (pmax > m[j]) {
        const float alpha = __expf(m[j] - pmax);
        l[j] *= alpha;
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[j][e] *= alpha;
        m[j] = pmax;
      }
      p[j] = (s == -INFINITY) ? 0.0f : __expf(s - m[j]);
      l[j] += wave_reduce_sum((lane & 3) == 0 ? p[j] : 0.0f);
      if ((lane & 3) == 0) p_lds[wave][j][tok] = p[j];
    }
    // intra-wave LDS write->read ordering is handled by the compiler's
    // lgkmcnt tracking (same wave, same LDS object)

    // ---- V pass: b128 per lane; 4 full V rows per wave instruction ----
    const int tlimit = min(BS, ctx - b * BS);  // tokens valid in this block
#pragma unroll
    for (int t4 = 0; t4 < 4; ++t4) {
      const int t = vtok + 4 * t4;
      if (t >= tlimit) continue;  // stale-block data may be non-finite
      ushort8 v8 = *reinterpret_cast<const ushort8*>(
          vbase + t * HD + vlane * 8);
      float vf[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) vf[e] = bf16_to_f32(v8.v[e]);
#pragma unroll
      for (int j = 0; j < G; ++j) {
        const float pj = p_lds[wave][j][t];
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[j][e] = fmaf(pj, vf[e], acc[j][e]);
      }
    }
  }

  // fold the 4 token-subgroup partial sums (lanes l, l+16, l+32, l+48 hold
  // the same dims)
#pragma unroll
  for (int j = 0; j < G; ++j) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      acc[j][e] += __shfl_xor(acc[j][e], 16, WAVE_SIZE);
      acc[j][e] += __shfl_xor(acc[j][e], 32, WAVE_SIZE);
    }
  }

  // ---- cross-wave merge ----
#pragma unroll
  for (int j = 0; j < G; ++j) {
    if (lane == 0) {
      m_lds[wave][j] = m[j];
      l_lds[wave][j] = l[j];
    }
    if (vtok == 0) {
#pragma unroll
      for (int e = 0; e < 8; ++e)
        acc_lds[wave][j][vlane * 8 + e] = acc[j][e];
    }
  }
  __syncthreads();

  for (int i = tid; i < G * HD; i += NWAVES * WAVE_SIZE) {
    const int j = i / HD;
    const int d = i % HD;
    float M = -INFINITY;
#pragma unroll
    for (int w = 0; w < NWAVES; ++w) M = fmaxf(M, m_lds[w][j]);
    float L = 0.0f, o = 0.0f;
#pragma unroll
    for (int w = 0; w < NWAVES; ++w) {
      const float f = (m_lds[w][j] == -INFINITY) ? 0.0f : __expf(m_lds[w][j] - M);
      L = fmaf(f, l_lds[w][j], L);
      o = fmaf(f, acc_lds[w][j][d], o);
    }
    out[((int64_t)seq * nq + kvh * G + j) * HD + d] = f32_to_bf16(o / L);
  }
}

template <int G>
static void launch_decode(torch::Tensor& out, const torch::Tensor& q,
                          const torch::Tensor& kcache,
                          const torch::Tensor& vcache,
                          const torch::Tensor& block_table,
                          const torch::Tensor& ctx_lens, float scale,
                          hipStream_t stream) {
  const int B = q.size(0);
  const int nkv = kcache.size(1);
  constexpr int NWAVES = 4;
  dim3 grid(nkv, B);
  hipLaunchKernelGGL((paged_attn_decode_kernel<128, G, NWAVES>), grid,
                     dim3(NWAVES * WAVE_SIZE), 0, stream,
                     (uint16_t*)out.data_ptr(), (const uint16_t*)q.data_ptr(),
                     (const uint16_t*)kcache.data_ptr(),
                     (const uint16_t*)vcache.data_ptr(),
                     (const int32_t*)block_table.data_ptr(),
                     (const int32_t*)ctx_lens.data_ptr(),
                     (int)block_table.size(1), nkv, scale);
  HIP_CHECK_KERNEL();
}

void paged_attn_decode(torch::Tensor out, torch::Tensor q,
                       torch::Tensor kcache, torch::Tensor vcache,
                       torch::Tensor block_table, torch::Tensor ctx_lens,
                       double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(kcache.size(3) == 128, "only head_dim=128 is supported");
  TORCH_CHECK(kcache.size(2) == 16, "only block_size=16 is supported");
  TORCH_CHECK(block_table.dtype() == torch::kInt32 &&
              ctx_lens.dtype() == torch::kInt32);
  const int nkv = kcache.size(1);
  const int nq = q.size(1);
  const int G = nq / nkv;
  TORCH_CHECK(nq % nkv == 0, "n_q_heads must be a multiple of n_kv_heads");
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  switch (G) {
    case 1:
      launch_decode<1>(out, q, kcache, vcache, block_table, ctx_lens,
                       (float)scale, stream);
      break;
    case 2:
      launch_decode<2>(out, q, kcache, vcache, block_table, ctx_lens,
                       (float)scale, stream);
      break;
    case 4:
      launch_decode<4>(out, q, kcache, vcache, block_table, ctx_lens,
                       (float)scale, stream);
      break;
    case 8:
      launch_decode<8>(out, q, kcache, vcache, block_table, ctx_lens,
                       (float)scale, stream);
      break;
    default:
      TORCH_CHECK(false, "unsupported GQA group size: ", G);
  }
}

}  // namespace igw
