// Fused top-k / top-p / temperature sampling kernel for gfx950.
//
// Net-new engine scope (SURVEY.md §7 stage 3b). One workgroup per sequence,
// operating on the fp32 logits row (vocab ~128k):
//
//   1. block max M and expsum Z (softmax denominator) in one pass;
//   2. greedy path (temperature == 0): argmax with lowest-index tie break;
//   3. top-k threshold: 4-level radix select over the monotone uint32 key
//      image of the fp32 logit (256-bin LDS histogram per level);
//   4. top-p (nucleus) threshold: same radix walk but with exp-weighted
//      histograms, stopping when the kept probability mass reaches p * Z;
//   5. final inverse-CDF sample over the kept set in index order, with the
//      caller-supplied uniform u (host draws it from torch's RNG so runs are
//      reproducible under torch.manual_seed).
//
// The Python reference implementation (ops/torch_ref.py::sample_ref) follows
// exactly these semantics (value-threshold top-k keeps ties; nucleus set is
// the smallest value-threshold set with mass >= p; index-order inverse CDF),
// so GPU vs CPU tests compare token-for-token with fixed u.

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "common.h"

namespace igw {

DEVINLINE uint32_t f32_key(float x) {
  union {
    float f;
    uint32_t u;
  } v;
  v.f = x;
  return (v.u & 0x80000000u) ? ~v.u : (v.u | 0x80000000u);
}

template <int BLOCK>
__global__ __launch_bounds__(BLOCK) void sample_kernel(
    int32_t* __restrict__ out,            // [B]
    const float* __restrict__ logits,     // [B, V]
    const float* __restrict__ temperature,// [B]
    const int32_t* __restrict__ top_k,    // [B] (<=0 => disabled)
    const float* __restrict__ top_p,      // [B] (>=1 => disabled)
    const float* __restrict__ uniforms,   // [B]
    int V) {
  __shared__ float redf[BLOCK / WAVE_SIZE];
  __shared__ float fbins[256];
  __shared__ uint32_t ubins[256];
  __shared__ uint32_t s_u32[2];
  __shared__ float s_f32[2];
  __shared__ float seg_sums[BLOCK + 1];

  const int seq = blockIdx.x;
  const int tid = threadIdx.x;
  const float* row = logits + (int64_t)seq * V;
  const float temp = temperature[seq];
  const bool greedy = (temp <= 0.0f);
  const float inv_t = greedy ? 1.0f : 1.0f / temp;

  // ---- pass 1: max ----
  float mx = -INFINITY;
  for (int i = tid; i < V; i += BLOCK) mx = fmaxf(mx, row[i] * inv_t);
  {
    // block reduce max via LDS
    int lane = tid & 63, wave = tid >> 6;
    mx = wave_reduce_max(mx);
    if (lane == 0) redf[wave] = mx;
    __syncthreads();
    float r = (tid < BLOCK / WAVE_SIZE) ? redf[tid] : -INFINITY;
    for (int off = 1; off < BLOCK / WAVE_SIZE; off <<= 1)
      r = fmaxf(r, __shfl_xor(r, off, WAVE_SIZE));
    if (tid == 0) s_f32[0] = r;
    __syncthreads();
  }
  const float M = s_f32[0];

  if (greedy) {
    // argmax, lowest index wins ties
    if (tid == 0) s_u32[0] = 0xffffffffu;
    __syncthreads();
    for (int i = tid; i < V; i += BLOCK) {
      if (row[i] == M) atomicMin(&s_u32[0], (uint32_t)i);
    }
    __syncthreads();
    if (tid == 0) out[seq] = (int32_t)s_u32[0];
    return;
  }

  // ---- pass 2: expsum Z ----
  float z = 0.0f;
  for (int i = tid; i < V; i += BLOCK) z += __expf(row[i] * inv_t - M);
  z = block_reduce_sum<BLOCK>(z, redf);
  const float Z = z;

  const int k = top_k[seq];
  const float p = top_p[seq];

  // ---- top-k radix select (over uint32 keys, MSB byte first) ----
  uint32_t tk_key = 0;  // keep everything by default
  if (k > 0 && k < V) {
    uint32_t prefix = 0;
    uint32_t remaining = (uint32_t)k;
    for (int level = 3; level >= 0; --level) {
      for (int b = tid; b < 256; b += BLOCK) ubins[b] = 0;
      __syncthreads();
      for (int i = tid; i < V; i += BLOCK) {
        const uint32_t key = f32_key(row[i] * inv_t);
        const bool match =
            (level == 3) || ((key >> ((level + 1) * 8)) == prefix);
        if (match) atomicAdd(&ubins[(key >> (level * 8)) & 0xff], 1u);
      }
      __syncthreads();
      if (tid == 0) {
        uint32_t cum = 0;
        int bstar = 0;
        for (int b = 255; b >= 0; --b) {
          if (cum + ubins[b] >= remaining) {
            bstar = b;
            break;
          }
          cum += ubins[b];
        }
        s_u32[0] = (prefix << 8) | (uint32_t)bstar;
        s_u32[1] = remaining - cum;
      }
      __syncthreads();
      prefix = s_u32[0];
      remaining = s_u32[1];
    }
    tk_key = prefix;
  }

  // ---- top-p radix select (exp-weighted histogram) ----
  uint32_t tp_key = 0;
  if (p < 1.0f) {
    uint32_t prefix = 0;
    float target = p * Z;
    for (int level = 3; level >= 0; --level) {
      for (int b = tid; b < 256; b += BLOCK) fbins[b] = 0.0f;
      __syncthreads();
      for (int i = tid; i < V; i += BLOCK) {
        const float x = row[i] * inv_t;
        const uint32_t key = f32_key(x);
        const bool match =
            (level == 3) || ((key >> ((level + 1) * 8)) == prefix);
        if (match) atomicAdd(&fbins[(key >> (level * 8)) & 0xff], __expf(x - M));
      }
      __syncthreads();
      if (tid == 0) {
        float cum = 0.0f;
        int bstar = 0;
        for (int b = 255; b >= 0; --b) {
          if (cum + fbins[b] >= target) {
            bstar = b;
            break;
          }
          cum += fbins[b];
        }
        s_u32[0] = (prefix << 8) | (uint32_t)bstar;
        s_f32[1] = target - cum;
      }
      __syncthreads();
      prefix = s_u32[0];
      target = s_f32[1];
    }
    tp_key = prefix;
  }

  const uint32_t tau = max(tk_key, tp_key);

  // ---- kept-mass Zf ----
  float zf = 0.0f;
  for (int i = tid; i < V; i += BLOCK) {
    const float x = row[i] * inv_t;
    if (f32_key(x) >= tau) zf += __expf(x - M);
  }
  zf = block_reduce_sum<BLOCK>(zf, redf);

  // ---- inverse-CDF sample in index order ----
  const float target = uniforms[seq] * zf;
  const int seg = (V + BLOCK - 1) / BLOCK;
  const int s0 = tid * seg;
  const int s1 = min(V, s0 + seg);
  float segsum = 0.0f;
  for (int i = s0; i < s1; ++i) {
    const float x = row[i] * inv_t;
    if (f32_key(x) >= tau) segsum += __expf(x - M);
  }
  seg_sums[tid] = segsum;
  __syncthreads();
  // exclusive scan over BLOCK segment sums (single thread is fine: 256 adds)
  if (tid == 0) {
    float run = 0.0f;
    for (int t = 0; t < BLOCK; ++t) {
      const float s = seg_sums[t];
      seg_sums[t] = run;
      run += s;
    }
    seg_sums[BLOCK] = run;
    s_u32[0] = 0xffffffffu;
  }
  __syncthreads();
  // the thread whose segment contains the target walks it
  const float base = seg_sums[tid];
  const float next = seg_sums[tid + 1];
  if (target >= base && (target < next || (tid == BLOCK - 1))) {
    float run = base;
    int chosen = -1;
    for (int i = s0; i < s1; ++i) {
      const float x = row[i] * inv_t;
      if (f32_key(x) >= tau) {
        run += __expf(x - M);
        chosen = i;
        if (run > target) break;
      }
    }
    if (chosen >= 0) atomicMin(&s_u32[0], (uint32_t)chosen);
  }
  __syncthreads();
  if (tid == 0) {
    uint32_t c = s_u32[0];
    if (c == 0xffffffffu) {
      // numeric corner (u ~ 1.0 and rounding): fall back to last kept token
      // -> handled by scanning from the top thread; choose argmax instead.
      c = 0;
      float best = -INFINITY;
      for (int i = 0; i < V; ++i) {
        // cheap host-style fallback; practically unreachable
        const float x = row[i] * inv_t;
        if (x > best) {
          best = x;
          c = i;
        }
      }
    }
    out[seq] = (int32_t)c;
  }
}

void sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temperature,
            torch::Tensor top_k, torch::Tensor top_p, torch::Tensor uniforms) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kFloat32);
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(out.dtype() == torch::kInt32);
  const int B = logits.size(0);
  const int V = logits.size(1);
  constexpr int BLOCK = 256;
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((sample_kernel<BLOCK>), dim3(B), dim3(BLOCK), 0, stream,
                     (int32_t*)out.data_ptr(), (const float*)logits.data_ptr(),
                     (const float*)temperature.data_ptr(),
                     (const int32_t*)top_k.data_ptr(),
                     (const float*)top_p.data_ptr(),
                     (const float*)uniforms.data_ptr(), V);
  HIP_CHECK_KERNEL();
}

}  // namespace igw
