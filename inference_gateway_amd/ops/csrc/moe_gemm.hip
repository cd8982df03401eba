// Grouped MFMA GEMM for MoE experts on gfx950 (BASELINE config 5:
// "grouped GEMM + expert all-to-all").
//
// Computes, for every expert e with per-expert row count counts[e]:
//     OUT[e*C + r, n] = sum_k A[e*C + r, k] * W[e, n, k]
// over the padded dispatch buffer written by the moe_route kernel — but
// unlike the torch batched-GEMM path it only COMPUTES tiles whose row
// range intersects counts[e] (the padded bmm burned 4x the FLOPs at
// decode batch 256: profiles/r01).  Design (cdna_hip_programming.md §5
// "Minimum 2-phase" recipe):
//
//   * tile BM=64 x BN=128 x BK=64, 4 waves, each wave a 32x64 sub-tile of
//     16x16x32 bf16 MFMAs accumulating in AGPRs;
//   * A staged global->reg->LDS (T14 split, double buffered, XOR
//     swizzle on write and read — guide T2); W is NOT staged: each
//     expert's weight tile is streamed exactly once per workgroup and a
//     lane's 16 contiguous bytes of a W row ARE its MFMA B-fragment, so
//     W fragments load straight to VGPRs with non-temporal hints (the
//     guide's "operand streamed once per block" rule — an LDS round
//     trip for W was pure overhead: 1.9 TB/s vs 4+);
//   * grid.y = E * C/BM row-tiles; a tile whose row base is past
//     counts[e] exits immediately, and stores are masked to real rows.
//
// Wavefront-64, no CUDA-isms, no Triton.

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace igw {

namespace {

constexpr int BM = 64;
constexpr int BN = 128;
constexpr int BK = 64;
constexpr int NWAVES = 4;

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

// swizzled byte offset of 16-byte piece `piece` of row `row` in a
// [rows][BK] bf16 LDS image (128-B rows, 8 pieces per row)
DEVINLINE int swz(int row, int piece) {
  return row * (BK * 2) + ((piece ^ (row & 7)) << 4);
}

__global__ __launch_bounds__(NWAVES * WAVE_SIZE) void moe_gemm_kernel(
    uint16_t* __restrict__ out,        // [E*C, N] bf16
    const uint16_t* __restrict__ a,    // [E*C, K] bf16
    const uint16_t* __restrict__ w,    // [E, N, K] bf16
    const int32_t* __restrict__ counts,  // [E]
    int C, int N, int K) {
  const int ntile = blockIdx.x;        // N/BN tiles
  const int rtile = blockIdx.y;        // E*(C/BM) row tiles
  const int rt_per_e = C / BM;
  const int e = rtile / rt_per_e;
  const int row0 = (rtile % rt_per_e) * BM;
  const int cnt = counts[e];
  if (row0 >= cnt) return;             // fully padded tile: nothing to do

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // LDS: double-buffered A [BM][BK] image only (swizzled)
  __shared__ uint16_t lds[2 * BM * BK];
#define A_LDS(b) (lds + (b) * BM * BK)

  const uint16_t* a_base = a + ((int64_t)e * C + row0) * K;
  const uint16_t* w_base = w + ((int64_t)e * N + ntile * BN) * K;
  // wave (wm, wn): wm in {0,1} rows, wn in {0,1} cols -> 32x64 out/wave
  const int wm = wave >> 1;
  const int wn = wave & 1;

  // ---- A reg staging: 2 pieces/thread (8 KiB); 8 16-B pieces per
  //      128-B tile row ----
  ushort8 ra[2];
  auto load_tile = [&](int k0) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int p = i * 256 + tid;
      ra[i] = *reinterpret_cast<const ushort8*>(
          a_base + (int64_t)(p >> 3) * K + k0 + (p & 7) * 8);
    }
  };
  auto write_tile = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int p = i * 256 + tid;
      *reinterpret_cast<ushort8*>(
          (char*)A_LDS(buf) + swz(p >> 3, p & 7)) = ra[i];
    }
  };
  // W fragment rows for this wave (fixed across the K loop)
  const int frow = lane & 15;
  const int fk0 = (lane >> 4) * 8;         // k element offset within BK=32
  const uint16_t* wrow[4];
#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    wrow[ni] = w_base + (int64_t)((wave & 1) * 64 + ni * 16 + frow) * K;
  }
  bf16x8 wfrag[2][4];
  auto load_wfrags = [&](int t) {
    const int k0 = t * BK;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        wfrag[ks][ni] = __builtin_nontemporal_load(
            reinterpret_cast<const bf16x8*>(wrow[ni] + k0 + ks * 32 + fk0));
      }
  };

  f32x4 acc[2][4];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  load_tile(0);
  write_tile(0);
  load_wfrags(0);
  if (K > BK) load_tile(BK);
  __syncthreads();

  const int nk = K / BK;
  int cur = 0;
  const int fkp = fk0 >> 3;                  // k piece index (0..3)
  for (int t = 0; t < nk; ++t) {
    // write A tile t+1 (loads issued an iteration ago), then issue A
    // tile t+2's loads so HBM latency hides under this step's MFMAs
    if (t + 1 < nk) {
      write_tile(cur ^ 1);
      if (t + 2 < nk) load_tile((t + 2) * BK);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {         // two K=32 sub-steps
      bf16x8 afrag[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        const int r = wm * 32 + mi * 16 + frow;
        afrag[mi] = *reinterpret_cast<const bf16x8*>(
            (const char*)A_LDS(cur) + swz(r, ks * 4 + fkp));
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], wfrag[ks][ni], acc[mi][ni], 0, 0, 0);
        }
    }
    __builtin_amdgcn_s_setprio(0);
    if (t + 1 < nk) load_wfrags(t + 1);      // next step's W fragments
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: C/D frag layout for 16x16: col=lane&15,
  //      row=(lane>>4)*4+j; store masked to real rows ----
  const int valid_rows = cnt - row0;
  uint16_t* out_base = out + ((int64_t)e * C + row0) * N + ntile * BN;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int r = wm * 32 + mi * 16 + (lane >> 4) * 4 + j;
        const int c = wn * 64 + ni * 16 + (lane & 15);
        if (r < valid_rows) {
          out_base[(int64_t)r * N + c] = f32_to_bf16(acc[mi][ni][j]);
        }
      }
#undef A_LDS
}

}  // namespace

void moe_grouped_gemm(torch::Tensor out, torch::Tensor a, torch::Tensor w,
                      torch::Tensor counts, int64_t capacity) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kBFloat16);
  TORCH_CHECK(w.dtype() == torch::kBFloat16 && out.dtype() == torch::kBFloat16);
  TORCH_CHECK(a.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(counts.dtype() == torch::kInt32);
  const int E = w.size(0);
  const int N = w.size(1);
  const int K = w.size(2);
  const int C = (int)capacity;
  TORCH_CHECK(a.size(0) == (int64_t)E * C && a.size(1) == K);
  TORCH_CHECK(out.size(0) == (int64_t)E * C && out.size(1) == N);
  TORCH_CHECK(C % BM == 0, "capacity must be a multiple of ", BM);
  TORCH_CHECK(N % BN == 0, "N must be a multiple of ", BN);
  TORCH_CHECK(K % BK == 0, "K must be a multiple of ", BK);
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  dim3 grid(N / BN, E * (C / BM));
  hipLaunchKernelGGL(moe_gemm_kernel, grid, dim3(NWAVES * WAVE_SIZE), 0,
                     stream, (uint16_t*)out.data_ptr(),
                     (const uint16_t*)a.data_ptr(),
                     (const uint16_t*)w.data_ptr(),
                     (const int32_t*)counts.data_ptr(), C, N, K);
  HIP_CHECK_KERNEL();
}

}  // namespace igw
