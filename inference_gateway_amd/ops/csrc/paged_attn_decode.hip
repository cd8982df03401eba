// Paged-attention decode kernel for gfx950 (MI355X).
//
// Net-new engine scope (SURVEY.md §7 stage 3b): single-query attention over a
// paged KV cache, GQA-aware. One workgroup per (sequence, kv_head); the GQA
// group's G query heads ride along so each K/V byte is read exactly once per
// step. Decode attention is HBM-bound; the kernel is built around coalesced
// traffic and LATENCY HIDING (profiles/r01: the naive per-head loop was
// latency-bound at 2.5 TB/s, 8x the per-iteration roofline):
//
//   * cache layout [num_blocks, n_kv_heads, block_size=16, head_dim] gives a
//     contiguous 16x128 bf16 tile (4 KiB) per (block, head);
//   * K pass: 64 lanes x 64 B (4 lanes per token x 32-dim segment); all G
//     head dot-products are computed back-to-back so their shfl reduction
//     chains overlap (ILP across heads instead of serial per-head chains);
//   * token-axis reductions use 4 shfl_xor steps (offsets 4,8,16,32 — the
//     token index lives in lane bits 2..5), not a full 6-step wave reduce;
//   * V rows for the CURRENT block and K rows for the NEXT block are issued
//     before the softmax shfl chain, so HBM latency hides under it;
//   * P values reach the V-pass lanes via ds_bpermute (no LDS round trip);
//   * V pass: 16 lanes x 8 contiguous dims (b128 reads), 4 token subgroups,
//     partials folded with 2 shfl_xor at the end;
//   * waves stripe over cache blocks and merge via LDS once at the end
//     (flash-decoding style rescale).
//
// Wavefront-64 throughout; no CUDA-isms, no Triton.

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace igw {

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

// xor-reduce over the token axis (lane bits 2..5); values stay replicated
DEVINLINE float token_reduce_max(float x) {
  x = fmaxf(x, __shfl_xor(x, 4, WAVE_SIZE));
  x = fmaxf(x, __shfl_xor(x, 8, WAVE_SIZE));
  x = fmaxf(x, __shfl_xor(x, 16, WAVE_SIZE));
  x = fmaxf(x, __shfl_xor(x, 32, WAVE_SIZE));
  return x;
}

DEVINLINE float token_reduce_sum(float x) {
  x += __shfl_xor(x, 4, WAVE_SIZE);
  x += __shfl_xor(x, 8, WAVE_SIZE);
  x += __shfl_xor(x, 16, WAVE_SIZE);
  x += __shfl_xor(x, 32, WAVE_SIZE);
  return x;
}

template <int HD, int G, int NWAVES>
__global__ __launch_bounds__(NWAVES* WAVE_SIZE) void paged_attn_decode_kernel(
    uint16_t* __restrict__ out,          // [B, NQ, HD]
    const uint16_t* __restrict__ q,      // [B, NQ, HD]
    const uint16_t* __restrict__ kcache, // [NB, NKV, BS, HD]
    const uint16_t* __restrict__ vcache, // [NB, NKV, BS, HD]
    const int32_t* __restrict__ block_table, // [B, max_blocks]
    const int32_t* __restrict__ ctx_lens,    // [B]
    int max_blocks, int nkv, float scale, int q_rs) {
  // q_rs: q row stride in elements (q may be a view into the fused QKV
  // GEMM output — no .contiguous() copy on the decode hot path)
  constexpr int BS = 16;  // tokens per cache block
  const int kvh = blockIdx.x;
  const int seq = blockIdx.y;
  const int nq = nkv * G;
  const int ctx = ctx_lens[seq];
  const int nblocks = (ctx + BS - 1) / BS;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE_SIZE - 1);
  const int wave = tid >> 6;

  __shared__ uint16_t q_lds[G][HD];
  __shared__ float m_lds[NWAVES][G];
  __shared__ float l_lds[NWAVES][G];
  __shared__ float acc_lds[NWAVES][G][HD];

  // stage Q into LDS (raw bf16; scale is applied to the scores so Q keeps
  // full precision and the per-lane register image stays PACKED — an
  // unpacked fp32 Q image is 128 VGPRs for G=4 and collapses occupancy)
  for (int i = tid * 8; i < G * HD; i += NWAVES * WAVE_SIZE * 8) {
    *reinterpret_cast<ushort8*>(&q_lds[0][0] + i) =
        *reinterpret_cast<const ushort8*>(
            q + (int64_t)seq * q_rs + (kvh * G) * HD + i);
  }
  __syncthreads();

  const int tok = lane >> 2;        // K pass: 16 tokens per tile
  const int seg = (lane & 3) * 32;  // 32-dim segment for the K dot
  const int vlane = lane & 15;      // V pass: dims [vlane*8, vlane*8+8)
  const int vtok = lane >> 4;       // V pass: tokens t = vtok + 4*t4

  float m[G], l[G], acc[G][8];
#pragma unroll
  for (int j = 0; j < G; ++j) {
    m[j] = -INFINITY;
    l[j] = 0.0f;
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[j][e] = 0.0f;
  }

  const int64_t head_off = (int64_t)kvh * BS * HD;
  const int32_t* bt_row = block_table + (int64_t)seq * max_blocks;

  // raw (unconverted) K tile registers for the current block
  ushort8 kraw[4];

  auto load_k = [&](int b, ushort8* dst) {
    const uint16_t* kp = kcache + ((int64_t)bt_row[b] * nkv) * BS * HD +
                         head_off + tok * HD + seg;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      dst[c] = *reinterpret_cast<const ushort8*>(kp + c * 8);
  };

  const int b0 = wave;
  if (b0 < nblocks) load_k(b0, kraw);

  for (int b = b0; b < nblocks; b += NWAVES) {
    // ---- scores for all G heads via v_dot2_f32_bf16: Q and K stay
    //      PACKED (no bf16 unpack VALU — v3's unpack loop made the
    //      kernel VALU-bound).  Q is re-read from LDS per head per block
    //      with the head loop NOT unrolled and the source pointer made
    //      opaque: a persistent (or unroll-duplicated) packed-Q image
    //      costs 48-64 VGPRs and pins the kernel at 2 waves/SIMD —
    //      occupancy (PMC: 60% of wave cycles parked) is the limiter,
    //      not the extra ds_reads ----
    float s[G];
    const bf16x2* kp2 = reinterpret_cast<const bf16x2*>(&kraw[0]);
#pragma unroll 1
    for (int j = 0; j < G; ++j) {
      const uint16_t* qsrc = &q_lds[j][seg];
      asm("" : "+v"(qsrc));
      uint32_t qtmp[16];
#pragma unroll
      for (int c4 = 0; c4 < 4; ++c4) {
        *reinterpret_cast<uint4*>(&qtmp[c4 * 4]) =
            *reinterpret_cast<const uint4*>(qsrc + c4 * 8);
      }
      const bf16x2* qp2 = reinterpret_cast<const bf16x2*>(&qtmp[0]);
      float sj = 0.0f;
#pragma unroll
      for (int c = 0; c < 16; ++c) {
        sj = __builtin_amdgcn_fdot2_f32_bf16(qp2[c], kp2[c], sj, false);
      }
      s[j] = sj;
    }

    // ---- issue V loads (current block) and K loads (next block) now so
    //      their latency hides under the shfl chains below; the scores
    //      consumed kraw above, so the next-K prefetch gets the whole
    //      softmax + V-accumulate phase of flight time ----
    const int tlimit = min(BS, ctx - b * BS);
    ushort8 vraw[4];
    const uint16_t* vp = vcache + ((int64_t)bt_row[b] * nkv) * BS * HD +
                         head_off + vlane * 8;
    if (tlimit == BS) {
#pragma unroll
      for (int t4 = 0; t4 < 4; ++t4)
        vraw[t4] = *reinterpret_cast<const ushort8*>(
            vp + (vtok + 4 * t4) * HD);
    } else {
#pragma unroll
      for (int t4 = 0; t4 < 4; ++t4)
        if (vtok + 4 * t4 < tlimit)
          vraw[t4] = *reinterpret_cast<const ushort8*>(
              vp + (vtok + 4 * t4) * HD);
    }
    if (b + NWAVES < nblocks) load_k(b + NWAVES, kraw);

    // ---- finish scores: pairwise dot reduce + mask ----
    const bool valid = (b * BS + tok) < ctx;
#pragma unroll
    for (int j = 0; j < G; ++j) {
      float v = s[j] + __shfl_xor(s[j], 1, WAVE_SIZE);
      v += __shfl_xor(v, 2, WAVE_SIZE);
      s[j] = valid ? v * scale : -INFINITY;
    }

    // ---- online softmax (all-head ILP) ----
    float pmax[G], p[G];
#pragma unroll
    for (int j = 0; j < G; ++j) pmax[j] = token_reduce_max(s[j]);
#pragma unroll
    for (int j = 0; j < G; ++j) {
      if (pmax[j] > m[j]) {
        const float alpha = __expf(m[j] - pmax[j]);
        l[j] *= alpha;
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[j][e] *= alpha;
        m[j] = pmax[j];
      }
      p[j] = (s[j] == -INFINITY) ? 0.0f : __expf(s[j] - m[j]);
    }
#pragma unroll
    for (int j = 0; j < G; ++j)
      l[j] += token_reduce_sum((lane & 3) == 0 ? p[j] : 0.0f);

    // ---- V accumulate: p reaches the V lanes via bpermute; math runs
    //      on float2 pairs (v_pk_fma_f32: 2 FLOPs/instr — no MFMAs here
    //      to contend with, so packed f32 is a straight 2x on the VALU
    //      issue count of this phase) ----
#pragma unroll
    for (int t4 = 0; t4 < 4; ++t4) {
      const int t = vtok + 4 * t4;
      float pj[G];
#pragma unroll
      for (int j = 0; j < G; ++j)
        pj[j] = __shfl(p[j], 4 * t, WAVE_SIZE);
      if (t >= tlimit) continue;
      const uint32_t* vp32 = reinterpret_cast<const uint32_t*>(&vraw[t4]);
      float2 vf2[4];
#pragma unroll
      for (int e2 = 0; e2 < 4; ++e2) {
        const uint32_t q = vp32[e2];
        vf2[e2].x = bf16_to_f32((uint16_t)(q & 0xffff));
        vf2[e2].y = bf16_to_f32((uint16_t)(q >> 16));
      }
#pragma unroll
      for (int j = 0; j < G; ++j) {
        float2* a2 = reinterpret_cast<float2*>(&acc[j][0]);
#pragma unroll
        for (int e2 = 0; e2 < 4; ++e2) {
          a2[e2].x = fmaf(pj[j], vf2[e2].x, a2[e2].x);
          a2[e2].y = fmaf(pj[j], vf2[e2].y, a2[e2].y);
        }
      }
    }

  }

  // fold the 4 token-subgroup partials (lanes l, l+16, l+32, l+48 share dims)
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
      const float f =
          (m_lds[w][j] == -INFINITY) ? 0.0f : __expf(m_lds[w][j] - M);
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
                     (int)block_table.size(1), nkv, scale,
                     (int)q.stride(0));
  HIP_CHECK_KERNEL();
}

void paged_attn_decode(torch::Tensor out, torch::Tensor q,
                       torch::Tensor kcache, torch::Tensor vcache,
                       torch::Tensor block_table, torch::Tensor ctx_lens,
                       double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(q.stride(-1) == 1 && q.stride(1) == q.size(2));
  TORCH_CHECK(out.is_contiguous());
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
