// Varlen causal flash-attention prefill kernel for gfx950 (MI355X).
//
// Net-new engine scope (SURVEY.md §7 stage 3b): prefill attention over the
// freshly-computed dense K/V of each prompt (the whole prompt is prefetched
// in one pass, so no paged-cache reads are needed here; the cache append is
// a separate fused scatter). MFMA-based, wavefront-64:
//
//   * mfma_f32_16x16x32_bf16 for both QK^T and PV;
//   * tile: BM=64 query rows x BN=64 keys per step, head_dim 128, 4 waves,
//     each wave owns 16 query rows;
//   * Q, K tiles staged in LDS with a (row&15)<<4 XOR byte-swizzle so the
//     16-lane ds_read_b128 fragment reads are bank-conflict-free (guide T2);
//   * V staged TRANSPOSED in LDS ([dim][key]) so the PV B-fragment reads 8
//     contiguous keys per lane; P round-trips through a per-wave LDS tile to
//     convert the C-layout scores into A-layout operands;
//   * online softmax in fp32 registers (per-row m/l live in the 16 lanes of
//     each C-column group).
//
// GQA: grid.y is the q head; the kv head is qh / (NQ/NKV).
// Varlen: host passes a flat tile map (tile -> sequence, tile offset) plus
// cu_seqlens, exactly one workgroup per (q-tile, q-head).

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace igw {

typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;

// byte-offset swizzles (XOR within a row; see guide T2 / rule 21)
DEVINLINE int swz256(int row, int colbyte) { return colbyte ^ ((row & 15) << 4); }
DEVINLINE int swz128(int row, int colbyte) { return colbyte ^ ((row & 7) << 4); }

template <int HD>
__global__ __launch_bounds__(256) void flash_prefill_kernel(
    uint16_t* __restrict__ out,       // [T, NQ, HD]
    const uint16_t* __restrict__ q,   // [T, NQ, HD]
    const uint16_t* __restrict__ k,   // [T, NKV, HD]
    const uint16_t* __restrict__ v,   // [T, NKV, HD]
    const int32_t* __restrict__ cu_seqlens,  // [B+1]
    const int32_t* __restrict__ tile_seq,    // [ntiles]
    const int32_t* __restrict__ tile_off,    // [ntiles] (q row offset in seq)
    int nq, int nkv, float scale, int q_rs, int k_rs, int v_rs) {
  // *_rs: token-row strides in elements — q/k/v may be views into the
  // fused QKV GEMM output (no .contiguous() copies)
  constexpr int BM = 64;
  constexpr int BN = 64;
  constexpr int NWAVES = 4;
  // LDS plan (bytes): Q[64][128]b16 @0 (16K), K[64][128]b16 @16K,
  // VT[128][64]b16 @32K (16K), P per-wave [16][64]b16 @48K (8K) = 56K
  __shared__ uint16_t smem[28 * 1024];
  uint16_t* q_s = smem;                  // swz256, row stride 128 elems
  uint16_t* k_s = smem + 8 * 1024;       // swz256
  uint16_t* vt_s = smem + 16 * 1024;     // swz128, row stride 64 elems
  uint16_t* p_s = smem + 24 * 1024;      // per wave: 16*64 elems

  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (nq / nkv);
  const int seq = tile_seq[tile];
  const int q0 = cu_seqlens[seq];        // first token of seq
  const int L = cu_seqlens[seq + 1] - q0;
  const int qoff = tile_off[tile];       // q position of tile row 0

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  uint16_t* pw_s = p_s + wave * 16 * 64;

  // ---- stage Q tile (rows qoff..qoff+63, clamped) ----
  {
    for (int it = 0; it < BM * HD / 8 / 256; ++it) {
      const int idx = tid + it * 256;
      const int row = idx / (HD / 8);
      const int d8 = (idx % (HD / 8)) * 8;
      const int grow = min(qoff + row, L - 1);
      ushort8 val = *reinterpret_cast<const ushort8*>(
          q + (int64_t)(q0 + grow) * q_rs + qh * HD + d8);
      *reinterpret_cast<ushort8*>(
          reinterpret_cast<char*>(q_s) + row * (HD * 2) +
          swz256(row, d8 * 2)) = val;
    }
  }

  // per-lane fragment indices
  const int frag_m = lane & 15;          // row-in-wave for A / col for B
  const int frag_k8 = (lane >> 4) * 8;   // k-offset of this lane's 8 elems
  const int crow0 = (lane >> 4) * 4;     // C/D: first of 4 rows this lane holds
  const int ccol = lane & 15;            // C/D: column this lane holds

  // Q fragments are loop-invariant: load once after the Q barrier.
  __syncthreads();
  short8v qfrag[HD / 32];
  {
    const int row = wave * 16 + frag_m;
#pragma unroll
    for (int kk = 0; kk < HD / 32; ++kk) {
      qfrag[kk] = *reinterpret_cast<const short8v*>(
          reinterpret_cast<char*>(q_s) + row * (HD * 2) +
          swz256(row, (kk * 32 + frag_k8) * 2));
    }
  }

  float m[4], l[4];
  float4v acc_o[HD / 16];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m[r] = -INFINITY;
    l[r] = 0.0f;
  }
#pragma unroll
  for (int nt = 0; nt < HD / 16; ++nt) acc_o[nt] = {0.f, 0.f, 0.f, 0.f};

  // causal: keys up to (last q row of tile, exclusive end)
  const int kend = min(L, qoff + BM);
  const int nkt = (kend + BN - 1) / BN;

  for (int kt = 0; kt < nkt; ++kt) {
    const int kbase = kt * BN;
    __syncthreads();  // previous tile fully consumed before restaging
    // ---- stage K tile ----
    for (int it = 0; it < BN * HD / 8 / 256; ++it) {
      const int idx = tid + it * 256;
      const int row = idx / (HD / 8);
      const int d8 = (idx % (HD / 8)) * 8;
      const int grow = min(kbase + row, L - 1);
      ushort8 val = *reinterpret_cast<const ushort8*>(
          k + (int64_t)(q0 + grow) * k_rs + kvh * HD + d8);
      *reinterpret_cast<ushort8*>(
          reinterpret_cast<char*>(k_s) + row * (HD * 2) +
          swz256(row, d8 * 2)) = val;
    }
    // ---- stage V transposed: VT[dim][key] ----
    for (int it = 0; it < BN * HD / 8 / 256; ++it) {
      const int idx = tid + it * 256;
      const int key = idx / (HD / 8);
      const int d8 = (idx % (HD / 8)) * 8;
      const int grow = min(kbase + key, L - 1);
      ushort8 val = *reinterpret_cast<const ushort8*>(
          v + (int64_t)(q0 + grow) * v_rs + kvh * HD + d8);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        *reinterpret_cast<uint16_t*>(
            reinterpret_cast<char*>(vt_s) + (d8 + e) * (BN * 2) +
            swz128(d8 + e, key * 2)) = val.v[e];
      }
    }
    __syncthreads();

    // ---- QK^T: 4 key sub-tiles x (HD/32) k-steps ----
    float4v acc_s[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      acc_s[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < HD / 32; ++kk) {
        const int krow = nt * 16 + frag_m;
        short8v bfrag = *reinterpret_cast<const short8v*>(
            reinterpret_cast<char*>(k_s) + krow * (HD * 2) +
            swz256(krow, (kk * 32 + frag_k8) * 2));
        acc_s[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], bfrag,
                                                            acc_s[nt], 0, 0, 0);
      }
    }

    // ---- mask + online softmax (per C row r; row data lives in the 16
    //      lanes sharing lane>>4) ----
    float pmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qpos = qoff + wave * 16 + crow0 + r;
      float mx = -INFINITY;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int kpos = kbase + nt * 16 + ccol;
        float s = acc_s[nt][r] * scale;
        if (kpos > qpos || kpos >= L) s = -INFINITY;
        acc_s[nt][r] = s;
        mx = fmaxf(mx, s);
      }
      pmax[r] = group_reduce_max<16>(mx);
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float mn = fmaxf(m[r], pmax[r]);
      alpha[r] = (m[r] == -INFINITY) ? 0.0f : __expf(m[r] - mn);
      m[r] = mn;
      float rowsum = 0.0f;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const float s = acc_s[nt][r];
        const float p = (s == -INFINITY) ? 0.0f : __expf(s - m[r]);
        acc_s[nt][r] = p;
        rowsum += p;
        // store P (bf16) for the PV A-operand: row = crow0+r, key col
        const int prow = crow0 + r;
        *reinterpret_cast<uint16_t*>(
            reinterpret_cast<char*>(pw_s) + prow * (BN * 2) +
            swz128(prow, (nt * 16 + ccol) * 2)) = f32_to_bf16(p);
      }
      l[r] = l[r] * alpha[r] + group_reduce_sum<16>(rowsum);
    }
    // rescale O accumulator
#pragma unroll
    for (int nt = 0; nt < HD / 16; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[nt][r] *= alpha[r];
    }

    // ---- PV: O += P * V  (A = P from LDS, B = VT fragments) ----
#pragma unroll
    for (int kk2 = 0; kk2 < BN / 32; ++kk2) {
      short8v pfrag = *reinterpret_cast<const short8v*>(
          reinterpret_cast<char*>(pw_s) + frag_m * (BN * 2) +
          swz128(frag_m, (kk2 * 32 + frag_k8) * 2));
#pragma unroll
      for (int nt = 0; nt < HD / 16; ++nt) {
        const int vrow = nt * 16 + frag_m;
        short8v vfrag = *reinterpret_cast<const short8v*>(
            reinterpret_cast<char*>(vt_s) + vrow * (BN * 2) +
            swz128(vrow, (kk2 * 32 + frag_k8) * 2));
        acc_o[nt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, acc_o[nt],
                                                    0, 0, 0);
      }
    }
  }

  // ---- epilogue: normalize and store ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = qoff + wave * 16 + crow0 + r;
    if (row >= L) continue;
    const float inv = 1.0f / l[r];
#pragma unroll
    for (int nt = 0; nt < HD / 16; ++nt) {
      out[((int64_t)(q0 + row) * nq + qh) * HD + nt * 16 + ccol] =
          f32_to_bf16(acc_o[nt][r] * inv);
    }
  }
}

// ---------------------------------------------------------------------------
// Paged-EXTEND prefill: queries are a chunk whose K/V (and the P cached
// prefix tokens before them) live in the PAGED cache — the enabler for
// chunked prefill and cross-request prefix caching (profiles/r02 lever 9).
// Same MFMA/LDS tiling as flash_prefill_kernel; only the K/V staging reads
// go through the block table, and the causal mask uses ABSOLUTE positions:
// q row r of a sequence with prefix P sits at position P + r and attends
// to cache positions [0, P + r].  The chunk's own K/V must already be
// appended to the cache (append_kv_cache runs first in the model).
// ---------------------------------------------------------------------------
template <int HD>
__global__ __launch_bounds__(256) void flash_prefill_paged_kernel(
    uint16_t* __restrict__ out,            // [Tq, NQ, HD]
    const uint16_t* __restrict__ q,        // [Tq, NQ, HD] (strided rows ok)
    const uint16_t* __restrict__ kcache,   // [NB, NKV, 16, HD]
    const uint16_t* __restrict__ vcache,   // [NB, NKV, 16, HD]
    const int32_t* __restrict__ block_table,  // [B, maxb]
    const int32_t* __restrict__ prefix_lens,  // [B] cached tokens before q
    const int32_t* __restrict__ cu_seqlens,   // [B+1] over q rows
    const int32_t* __restrict__ tile_seq, const int32_t* __restrict__ tile_off,
    int nq, int nkv, float scale, int q_rs, int max_blocks) {
  constexpr int BM = 64;
  constexpr int BN = 64;
  constexpr int BS = 16;                   // cache block size (tokens)
  __shared__ uint16_t smem[28 * 1024];
  uint16_t* q_s = smem;
  uint16_t* k_s = smem + 8 * 1024;
  uint16_t* vt_s = smem + 16 * 1024;
  uint16_t* p_s = smem + 24 * 1024;

  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (nq / nkv);
  const int seq = tile_seq[tile];
  const int q0 = cu_seqlens[seq];
  const int Lq = cu_seqlens[seq + 1] - q0;
  const int P = prefix_lens[seq];
  const int KVL = P + Lq;                  // total cached positions
  const int qoff = tile_off[tile];
  const int32_t* bt_row = block_table + (int64_t)seq * max_blocks;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  uint16_t* pw_s = p_s + wave * 16 * 64;

  for (int it = 0; it < BM * HD / 8 / 256; ++it) {
    const int idx = tid + it * 256;
    const int row = idx / (HD / 8);
    const int d8 = (idx % (HD / 8)) * 8;
    const int grow = min(qoff + row, Lq - 1);
    ushort8 val = *reinterpret_cast<const ushort8*>(
        q + (int64_t)(q0 + grow) * q_rs + qh * HD + d8);
    *reinterpret_cast<ushort8*>(reinterpret_cast<char*>(q_s) +
                                row * (HD * 2) + swz256(row, d8 * 2)) = val;
  }

  const int frag_m = lane & 15;
  const int frag_k8 = (lane >> 4) * 8;
  const int crow0 = (lane >> 4) * 4;
  const int ccol = lane & 15;

  __syncthreads();
  short8v qfrag[HD / 32];
  {
    const int row = wave * 16 + frag_m;
#pragma unroll
    for (int kk = 0; kk < HD / 32; ++kk) {
      qfrag[kk] = *reinterpret_cast<const short8v*>(
          reinterpret_cast<char*>(q_s) + row * (HD * 2) +
          swz256(row, (kk * 32 + frag_k8) * 2));
    }
  }

  float m[4], l[4];
  float4v acc_o[HD / 16];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m[r] = -INFINITY;
    l[r] = 0.0f;
  }
#pragma unroll
  for (int nt = 0; nt < HD / 16; ++nt) acc_o[nt] = {0.f, 0.f, 0.f, 0.f};

  // causal over ABSOLUTE positions: this tile's last q row sits at
  // P + qoff + BM - 1, so keys [0, min(KVL, P + qoff + BM)) participate
  const int kend = min(KVL, P + qoff + BM);
  const int nkt = (kend + BN - 1) / BN;

  for (int kt = 0; kt < nkt; ++kt) {
    const int kbase = kt * BN;
    __syncthreads();
    // ---- stage K tile from the paged cache ----
    for (int it = 0; it < BN * HD / 8 / 256; ++it) {
      const int idx = tid + it * 256;
      const int row = idx / (HD / 8);
      const int d8 = (idx % (HD / 8)) * 8;
      const int gpos = min(kbase + row, KVL - 1);
      const int64_t blk = bt_row[gpos / BS];
      ushort8 val = *reinterpret_cast<const ushort8*>(
          kcache + ((blk * nkv + kvh) * BS + (gpos % BS)) * HD + d8);
      *reinterpret_cast<ushort8*>(reinterpret_cast<char*>(k_s) +
                                  row * (HD * 2) + swz256(row, d8 * 2)) = val;
    }
    // ---- stage V transposed ----
    for (int it = 0; it < BN * HD / 8 / 256; ++it) {
      const int idx = tid + it * 256;
      const int key = idx / (HD / 8);
      const int d8 = (idx % (HD / 8)) * 8;
      const int gpos = min(kbase + key, KVL - 1);
      const int64_t blk = bt_row[gpos / BS];
      ushort8 val = *reinterpret_cast<const ushort8*>(
          vcache + ((blk * nkv + kvh) * BS + (gpos % BS)) * HD + d8);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        *reinterpret_cast<uint16_t*>(
            reinterpret_cast<char*>(vt_s) + (d8 + e) * (BN * 2) +
            swz128(d8 + e, key * 2)) = val.v[e];
      }
    }
    __syncthreads();

    float4v acc_s[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      acc_s[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < HD / 32; ++kk) {
        const int krow = nt * 16 + frag_m;
        short8v bfrag = *reinterpret_cast<const short8v*>(
            reinterpret_cast<char*>(k_s) + krow * (HD * 2) +
            swz256(krow, (kk * 32 + frag_k8) * 2));
        acc_s[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], bfrag,
                                                            acc_s[nt], 0, 0, 0);
      }
    }

    float pmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qpos_abs = P + qoff + wave * 16 + crow0 + r;
      float mx = -INFINITY;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int kpos = kbase + nt * 16 + ccol;
        float s = acc_s[nt][r] * scale;
        if (kpos > qpos_abs || kpos >= KVL) s = -INFINITY;
        acc_s[nt][r] = s;
        mx = fmaxf(mx, s);
      }
      pmax[r] = group_reduce_max<16>(mx);
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float mn = fmaxf(m[r], pmax[r]);
      alpha[r] = (m[r] == -INFINITY) ? 0.0f : __expf(m[r] - mn);
      m[r] = mn;
      float rowsum = 0.0f;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const float s = acc_s[nt][r];
        const float p = (s == -INFINITY) ? 0.0f : __expf(s - m[r]);
        acc_s[nt][r] = p;
        rowsum += p;
        const int prow = crow0 + r;
        *reinterpret_cast<uint16_t*>(
            reinterpret_cast<char*>(pw_s) + prow * (BN * 2) +
            swz128(prow, (nt * 16 + ccol) * 2)) = f32_to_bf16(p);
      }
      l[r] = l[r] * alpha[r] + group_reduce_sum<16>(rowsum);
    }
#pragma unroll
    for (int nt = 0; nt < HD / 16; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[nt][r] *= alpha[r];
    }

#pragma unroll
    for (int kk2 = 0; kk2 < BN / 32; ++kk2) {
      short8v pfrag = *reinterpret_cast<const short8v*>(
          reinterpret_cast<char*>(pw_s) + frag_m * (BN * 2) +
          swz128(frag_m, (kk2 * 32 + frag_k8) * 2));
#pragma unroll
      for (int nt = 0; nt < HD / 16; ++nt) {
        const int vrow = nt * 16 + frag_m;
        short8v vfrag = *reinterpret_cast<const short8v*>(
            reinterpret_cast<char*>(vt_s) + vrow * (BN * 2) +
            swz128(vrow, (kk2 * 32 + frag_k8) * 2));
        acc_o[nt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, acc_o[nt],
                                                    0, 0, 0);
      }
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = qoff + wave * 16 + crow0 + r;
    if (row >= Lq) continue;
    const float inv = 1.0f / l[r];
#pragma unroll
    for (int nt = 0; nt < HD / 16; ++nt) {
      out[((int64_t)(q0 + row) * nq + qh) * HD + nt * 16 + ccol] =
          f32_to_bf16(acc_o[nt][r] * inv);
    }
  }
}

void flash_prefill_paged(torch::Tensor out, torch::Tensor q,
                         torch::Tensor kcache, torch::Tensor vcache,
                         torch::Tensor block_table, torch::Tensor prefix_lens,
                         torch::Tensor cu_seqlens, torch::Tensor tile_seq,
                         torch::Tensor tile_off, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(q.stride(-1) == 1 && q.stride(1) == q.size(2));
  TORCH_CHECK(kcache.is_contiguous() && vcache.is_contiguous());
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(block_table.dtype() == torch::kInt32 &&
              block_table.is_contiguous());
  TORCH_CHECK(prefix_lens.dtype() == torch::kInt32);
  TORCH_CHECK(cu_seqlens.dtype() == torch::kInt32);
  TORCH_CHECK(kcache.size(2) == 16, "cache block size must be 16");
  const int HD = q.size(2);
  TORCH_CHECK(HD == 128, "only head_dim=128 supported");
  const int nq = q.size(1);
  const int nkv = kcache.size(1);
  TORCH_CHECK(nq % nkv == 0);
  const int ntiles = tile_seq.size(0);
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  dim3 grid(ntiles, nq);
  hipLaunchKernelGGL((flash_prefill_paged_kernel<128>), grid, dim3(256), 0,
                     stream, (uint16_t*)out.data_ptr(),
                     (const uint16_t*)q.data_ptr(),
                     (const uint16_t*)kcache.data_ptr(),
                     (const uint16_t*)vcache.data_ptr(),
                     (const int32_t*)block_table.data_ptr(),
                     (const int32_t*)prefix_lens.data_ptr(),
                     (const int32_t*)cu_seqlens.data_ptr(),
                     (const int32_t*)tile_seq.data_ptr(),
                     (const int32_t*)tile_off.data_ptr(), nq, nkv,
                     (float)scale, (int)q.stride(0),
                     (int)block_table.size(1));
  HIP_CHECK_KERNEL();
}

void flash_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                   torch::Tensor v, torch::Tensor cu_seqlens,
                   torch::Tensor tile_seq, torch::Tensor tile_off,
                   double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  for (auto* t : {&q, &k, &v}) {
    TORCH_CHECK(t->stride(-1) == 1 && t->stride(1) == t->size(2));
  }
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(cu_seqlens.dtype() == torch::kInt32);
  const int HD = q.size(2);
  TORCH_CHECK(HD == 128, "only head_dim=128 supported");
  const int nq = q.size(1);
  const int nkv = k.size(1);
  TORCH_CHECK(nq % nkv == 0);
  const int ntiles = tile_seq.size(0);
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  dim3 grid(ntiles, nq);
  hipLaunchKernelGGL((flash_prefill_kernel<128>), grid, dim3(256), 0, stream,
                     (uint16_t*)out.data_ptr(), (const uint16_t*)q.data_ptr(),
                     (const uint16_t*)k.data_ptr(),
                     (const uint16_t*)v.data_ptr(),
                     (const int32_t*)cu_seqlens.data_ptr(),
                     (const int32_t*)tile_seq.data_ptr(),
                     (const int32_t*)tile_off.data_ptr(), nq, nkv,
                     (float)scale, (int)q.stride(0), (int)k.stride(0),
                     (int)v.stride(0));
  HIP_CHECK_KERNEL();
}

}  // namespace igw
