// Fused TN decode GEMM v2: block-shared weight tiles through LDS.
//
// v1 (fused_gemm.hip) measured 0.2 TB/s — its dominant defect was W
// DUPLICATION: every wave of a block (and every M-tile block) streamed
// the same W fragments from global, multiplying the weight traffic by
// up to 32x.  v2 restructures per the guide's M=256 projection-GEMM
// recipe:
//
//   * one block covers BM=256 rows (8 waves x 32 rows) x BN=64 output
//     cols, so W is fetched once per block per k-step into LDS (glds,
//     double-buffered) and all 8 waves read their B fragments from LDS;
//   * grid = (N/64, M/256): only M/256 = 4 blocks ever touch a W line,
//     and they are XCD-colocated for L2 reuse because the dispatch id
//     stride between M-siblings is gridDim.x (224 for the flagship
//     shapes, ≡ 0 mod 8 — guide §Workgroup dispatch: block b lands on
//     XCD b%8, a speed-only observation);
//   * the LDS W image is XOR-swizzled (byte ^= (row&7)<<4) against the
//     16-way ds_read_b128 bank conflict of 128-B rows (guide §6 G4),
//     with the inverse swizzle applied on the glds SOURCE address
//     (rule 21: linear dest + pre-swizzled source + swizzled read);
//   * x (L2/LLC-resident, re-read by every N-tile) stays on direct
//     per-wave fragment loads;
//   * simple 2-phase k-loop (stage next | compute current, one
//     vmcnt(0)+barrier pair per BK=64 step) — the guide's minimum
//     pipelined form; the 8-phase counted-vmcnt schedule is the known
//     next step if this one beats the library.
//
// SWIGLU mode fuses silu(gate)*up exactly like v1.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace igw {

typedef __attribute__((ext_vector_type(8))) short bf16x8_2;
typedef __attribute__((ext_vector_type(4))) float f32x4_2;

constexpr int BM2 = 256;    // rows per block (8 waves x 32)
constexpr int BN2 = 64;     // output cols per block
constexpr int BK2 = 64;     // k per pipeline step

// debug probe: stage a [64][64]-element bf16 matrix through the same
// glds path (optionally source-swizzled) and read it back linearly —
// pins the glds lane-placement and swizzle-pairing assumptions.
__global__ void __launch_bounds__(512)
glds_probe_kernel(uint16_t* __restrict__ out,
                  const uint16_t* __restrict__ inp, int swizzle) {
  __shared__ uint8_t sm[8192];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int base = wave * 1024;
  const int o = base + lane * 16;
  const int row = o / 128;
  const int bir = o % 128;
  const int src_byte = swizzle ? (bir ^ ((row & 7) << 4)) : bir;
  const uint16_t* src = inp + row * 64 + src_byte / 2;
  __builtin_amdgcn_global_load_lds(
      reinterpret_cast<const unsigned int*>(src),
      reinterpret_cast<unsigned int*>(sm + base), 16, 0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  for (int i = threadIdx.x; i < 64 * 64; i += 512) {
    const int r = i / 64;
    const int e = i % 64;
    const int b = r * 128
        + ((e * 2) ^ (swizzle ? ((r & 7) << 4) : 0));
    out[i] = *reinterpret_cast<const uint16_t*>(sm + b);
  }
}

void glds_probe(torch::Tensor out, torch::Tensor inp, int64_t swizzle) {
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(glds_probe_kernel, dim3(1), dim3(512), 0, stream,
                     reinterpret_cast<uint16_t*>(out.data_ptr()),
                     reinterpret_cast<const uint16_t*>(inp.data_ptr()),
                     (int)swizzle);
}

template <bool SWIGLU, bool SWZ>
__global__ void __launch_bounds__(512)
fused_tn_gemm2_kernel(uint16_t* __restrict__ out,
                      const uint16_t* __restrict__ x,
                      const uint16_t* __restrict__ w,
                      int M, int N, int K, long x_row, long w_row,
                      long o_row) {
  constexpr int WROWS = SWIGLU ? 2 * BN2 : BN2;   // W rows per k-step
  constexpr int ROWB = BK2 * 2;                   // 128 bytes per row
  // [2 buffers][WROWS rows][128 bytes]
  __shared__ uint8_t smem[2][WROWS][ROWB];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int n_tile = blockIdx.x;
  const int m_tile = blockIdx.y;
  const int row0 = m_tile * BM2 + wave * 32;
  const int col0 = n_tile * BN2;
  const int fr = lane & 15;
  const int kg = lane >> 4;

  f32x4_2 acc_g[2][4] = {};
  f32x4_2 acc_u[2][4] = {};

  // ---- glds staging: WROWS*128 bytes per step.  One glds moves 1 KiB
  // per WAVE (wave-uniform LDS base; the hardware lands lane i at
  // base + i*16).  Linear LDS offset o = wave_piece_base + lane*16;
  // row = o/128, byte_in_row = o%128; the SOURCE byte is pre-swizzled
  // (rule 21) so the linear LDS image reads back swizzled.
  constexpr int PPW = (WROWS * ROWB) / (8 * 1024);  // pieces per wave
  const uint16_t* wg_base = w + (long)col0 * w_row;
  const uint16_t* wu_base = SWIGLU ? w + (long)(N + col0) * w_row
                                   : nullptr;

  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int p = 0; p < PPW; ++p) {
      const int base = (wave * PPW + p) * 1024;
      const int o = base + lane * 16;
      const int row = o / ROWB;
      const int byte_in_row = o % ROWB;
      const int src_byte = SWZ ? (byte_in_row ^ ((row & 7) << 4))
                                : byte_in_row;
      const uint16_t* src;
      if (!SWIGLU || row < BN2) {
        src = wg_base + (long)row * w_row + k0 + src_byte / 2;
      } else {
        src = wu_base + (long)(row - BN2) * w_row + k0 + src_byte / 2;
      }
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int*>(src),
          reinterpret_cast<unsigned int*>(&smem[buf][0][0] + base),
          16, 0, 0);
    }
  };

  // B fragment read: col j = fr within this n-frag, k-group kg
  auto ldb = [&](int buf, int base_row, int nfrag,
                 int khalf) -> bf16x8_2 {
    const int row = base_row + nfrag * 16 + fr;
    // a row holds BK2=64 elements (128 B); k-half h covers bytes
    // [h*64, h*64+64), k-group kg its 16-B slot within the half
    const int byte_in_row = khalf * 64 + kg * 16;
    const int b = row * ROWB
        + (SWZ ? (byte_in_row ^ ((row & 7) << 4)) : byte_in_row);
    return *reinterpret_cast<const bf16x8_2*>(&smem[buf][0][0] + b);
  };

  const uint16_t* xa0 = x + (long)(row0 + fr) * x_row + kg * 8;
  const uint16_t* xa1 = xa0 + 16 * x_row;

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const int nsteps = K / BK2;
  for (int t = 0; t < nsteps; ++t) {
    const int k0 = t * BK2;
    const int cur = t & 1;
    if (t + 1 < nsteps) {
      stage(cur ^ 1, k0 + BK2);
    }
    // A fragments for both k-halves of this step (direct global; the
    // x slab is L2-resident)
    bf16x8_2 a[2][2];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      a[0][h] = *reinterpret_cast<const bf16x8_2*>(xa0 + k0 + h * 32);
      a[1][h] = *reinterpret_cast<const bf16x8_2*>(xa1 + k0 + h * 32);
    }
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const bf16x8_2 bg = ldb(cur, 0, n, h);
        acc_g[0][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[0][h], bg, acc_g[0][n], 0, 0, 0);
        acc_g[1][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[1][h], bg, acc_g[1][n], 0, 0, 0);
        if (SWIGLU) {
          const bf16x8_2 bu = ldb(cur, BN2, n, h);
          acc_u[0][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[0][h], bu, acc_u[0][n], 0, 0, 0);
          acc_u[1][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[1][h], bu, acc_u[1][n], 0, 0, 0);
        }
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  // ---- epilogue (same D mapping as v1) ----
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int jj = 0; jj < 4; ++jj) {
        const int i = kg * 4 + jj;
        const int orow = row0 + m * 16 + i;
        const int ocol = col0 + n * 16 + fr;
        float v = acc_g[m][n][jj];
        if (SWIGLU) {
          const float g = v;
          const float s = g / (1.0f + __expf(-g));
          v = s * acc_u[m][n][jj];
        }
        out[(long)orow * o_row + ocol] = f32_to_bf16(v);
      }
    }
  }
}

void fused_gemm_tn2(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                    bool swiglu, bool swz) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda() && out.is_cuda());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.scalar_type() == torch::kBFloat16);
  const int M = x.size(0), K = x.size(1);
  const int N = swiglu ? w.size(0) / 2 : w.size(0);
  TORCH_CHECK(w.size(1) == K && out.size(0) == M && out.size(1) == N);
  TORCH_CHECK(M % BM2 == 0 && N % BN2 == 0 && K % BK2 == 0,
              "fused_gemm_tn2 needs M%256==0, N%64==0, K%64==0");
  TORCH_CHECK(x.stride(1) == 1 && w.stride(1) == 1
              && out.stride(1) == 1);
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(N / BN2, M / BM2);
  auto launch = [&](auto sg, auto sz) {
    hipLaunchKernelGGL(
        (fused_tn_gemm2_kernel<decltype(sg)::value,
                               decltype(sz)::value>),
        grid, dim3(512), 0, stream,
        reinterpret_cast<uint16_t*>(out.data_ptr()),
        reinterpret_cast<const uint16_t*>(x.data_ptr()),
        reinterpret_cast<const uint16_t*>(w.data_ptr()),
        M, N, K, x.stride(0), w.stride(0), out.stride(0));
  };
  using T = std::true_type;
  using F = std::false_type;
  if (swiglu) {
    swz ? launch(T{}, T{}) : launch(T{}, F{});
  } else {
    swz ? launch(F{}, T{}) : launch(F{}, F{});
  }
}

}  // namespace igw
