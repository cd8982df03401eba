// Fused top-k / top-p / temperature sampling kernel for gfx950.
//
// Net-new engine scope (SURVEY.md §7 stage 3b). One workgroup per sequence,
// operating on the fp32 logits row (vocab ~128k):
//
//   1. block max M in one pass;
//   2. greedy path (temperature == 0): argmax with lowest-index tie break;
//   3. joint 4-level radix walk over the monotone uint32 key image of the
//      fp32 logit: each level is ONE pass that builds a count histogram
//      (for the top-k threshold) and an exp-weighted histogram (for the
//      top-p mass threshold) simultaneously, into WAVE-PRIVATE LDS bins
//      (logit exponents cluster in a handful of bins, so shared-bin atomics
//      would serialize across the whole block); Z falls out of level 3's
//      weight histogram for free;
//   4. kept-mass pass, then an inverse-CDF walk over the kept set in index
//      order with the caller-supplied uniform u (host draws it from torch's
//      RNG so runs are reproducible under torch.manual_seed).
//
// Thresholds are VALUE thresholds: top-k keeps ties of the k-th value; the
// nucleus set is the smallest value-threshold set with mass >= p*Z. The
// Python reference (ops/torch_ref.py::sample_ref) implements exactly these
// semantics, so GPU vs CPU tests compare token-for-token with fixed u.

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
  constexpr int NW = BLOCK / WAVE_SIZE;
  __shared__ float redf[NW];
  __shared__ uint32_t ubins[NW][256];
  __shared__ float fbins[NW][256];
  __shared__ uint32_t s_u32[2];
  __shared__ float s_f32[4];
  __shared__ float seg_sums[BLOCK + 1];

  const int seq = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const float* row = logits + (int64_t)seq * V;
  const float temp = temperature[seq];
  const bool greedy = (temp <= 0.0f);
  const float inv_t = greedy ? 1.0f : 1.0f / temp;

  // ---- pass 1: max ----
  float mx = -INFINITY;
  for (int i = tid; i < V; i += BLOCK) mx = fmaxf(mx, row[i] * inv_t);
  mx = wave_reduce_max(mx);
  if (lane == 0) redf[wave] = mx;
  __syncthreads();
  if (tid == 0) {
    float r = redf[0];
    for (int w = 1; w < NW; ++w) r = fmaxf(r, redf[w]);
    s_f32[0] = r;
  }
  __syncthreads();
  const float M = s_f32[0];

  if (greedy) {
    if (tid == 0) s_u32[0] = 0xffffffffu;
    __syncthreads();
    for (int i = tid; i < V; i += BLOCK) {
      if (row[i] == M) atomicMin(&s_u32[0], (uint32_t)i);
    }
    __syncthreads();
    if (tid == 0) out[seq] = (int32_t)s_u32[0];
    return;
  }

  const int k = top_k[seq];
  const float p = top_p[seq];
  const bool do_k = (k > 0 && k < V);
  const bool do_p = (p < 1.0f);

  // ---- joint radix walk: 4 levels, one pass each ----
  uint32_t tk_prefix = 0, tp_prefix = 0;
  uint32_t tk_remaining = do_k ? (uint32_t)k : 0;
  float tp_target = 0.0f;  // set after level 3 (needs Z)
  float Z = 0.0f;

  for (int level = 3; level >= 0; --level) {
    for (int b = tid; b < 256; b += BLOCK) {
#pragma unroll
      for (int w = 0; w < NW; ++w) {
        ubins[w][b] = 0;
        fbins[w][b] = 0.0f;
      }
    }
    __syncthreads();
    const int shift = level * 8;
    for (int i = tid; i < V; i += BLOCK) {
      const float x = row[i] * inv_t;
      const uint32_t key = f32_key(x);
      const uint32_t hi = (level == 3) ? 0 : (key >> (shift + 8));
      const uint32_t b = (key >> shift) & 0xff;
      if (do_k && (level == 3 || hi == tk_prefix)) {
        atomicAdd(&ubins[wave][b], 1u);
      }
      if (level == 3 || hi == tp_prefix) {
        // level 3 weight histogram doubles as Z (do_p or not)
        atomicAdd(&fbins[wave][b], __expf(x - M));
      }
    }
    __syncthreads();
    if (tid == 0) {
      if (level == 3) {
        float z = 0.0f;
        for (int b = 0; b < 256; ++b)
          for (int w = 0; w < NW; ++w) z += fbins[w][b];
        Z = z;
        s_f32[1] = z;
        tp_target = p * z;
      }
      if (do_k) {
        uint32_t cum = 0;
        int bstar = 0;
        for (int b = 255; b >= 0; --b) {
          uint32_t c = 0;
          for (int w = 0; w < NW; ++w) c += ubins[w][b];
          if (cum + c >= tk_remaining) {
            bstar = b;
            break;
          }
          cum += c;
        }
        tk_prefix = (tk_prefix << 8) | (uint32_t)bstar;
        tk_remaining -= cum;
        s_u32[0] = tk_prefix;
      }
      if (do_p) {
        float cum = 0.0f;
        int bstar = 0;
        for (int b = 255; b >= 0; --b) {
          float c = 0.0f;
          for (int w = 0; w < NW; ++w) c += fbins[w][b];
          if (cum + c >= tp_target) {
            bstar = b;
            break;
          }
          cum += c;
        }
        tp_prefix = (tp_prefix << 8) | (uint32_t)bstar;
        tp_target -= cum;
        s_f32[2] = tp_target;
        s_u32[1] = tp_prefix;
      }
    }
    __syncthreads();
    if (do_k) tk_prefix = s_u32[0];
    if (do_p) {
      tp_prefix = s_u32[1];
      tp_target = s_f32[2];
    }
    // each thread needs tk_remaining only on thread 0; prefixes broadcast
    if (!do_k && !do_p && level == 3) break;  // pure-temperature sampling
  }

  const uint32_t tau =
      max(do_k ? tk_prefix : 0u, do_p ? tp_prefix : 0u);

  // ---- kept mass + per-thread segment sums (one fused pass) ----
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
  if (tid == 0) {
    float run = 0.0f;
    for (int t = 0; t < BLOCK; ++t) {
      const float s = seg_sums[t];
      seg_sums[t] = run;
      run += s;
    }
    seg_sums[BLOCK] = run;  // = Zf
    s_u32[0] = 0xffffffffu;
  }
  __syncthreads();

  // ---- inverse-CDF walk in index order ----
  const float zf = seg_sums[BLOCK];
  const float target = uniforms[seq] * zf;
  const float base = seg_sums[tid];
  const float next = seg_sums[tid + 1];
  if (target >= base && (target < next || tid == BLOCK - 1)) {
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
      // numeric corner (u ~ 1.0): fall back to argmax
      c = 0;
      float best = -INFINITY;
      for (int i = 0; i < V; ++i) {
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

// ---------------------------------------------------------------------------
// bf16 fast path.  The lm_head emits bf16; sampling directly on the bf16
// bits (a) halves the logits traffic vs the fp32 path and (b) shrinks the
// monotone key to 16 bits, so the radix walk is 2 levels instead of 4.
// Thresholds are found on the UNSCALED bf16 key (temperature > 0 preserves
// order); only the exp-weights use the temperature.  Loads are uint4
// (8 bf16/lane), and per-8 runs of equal bins are merged before the LDS
// atomic — with near-flat random-init logits the bins collide wave-wide,
// which is what made the fp32 kernel's histogram passes serialize
// (profiles/r01: 1.48 ms at B=512).
// ---------------------------------------------------------------------------

DEVINLINE uint32_t bf16_key(uint16_t u) {
  return (u & 0x8000u) ? (uint32_t)(~u & 0xffffu) : (uint32_t)(u | 0x8000u);
}

DEVINLINE float bf16_val(uint16_t u) {
  union {
    uint32_t q;
    float f;
  } v;
  v.q = ((uint32_t)u) << 16;
  return v.f;
}

// Parallel suffix-threshold search over 256 combined bins, run by ONE wave.
// vals: combined per-bin values (descending walk from bin 255).  Returns
// (bstar, remaining) through s_res: the first bin b (from the top) where
// cum_above + vals[b] >= target, with remaining = target - cum_above.
template <typename T>
DEVINLINE void suffix_threshold_search(const T* vals, T target, int lane,
                                       uint32_t* s_b, T* s_rem) {
  T lsum = T(0);
#pragma unroll
  for (int j = 0; j < 4; ++j) lsum += vals[lane * 4 + j];
  // inclusive suffix sum across lanes: s[l] = sum_{l' >= l} lsum[l']
  T s = lsum;
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) {
    const T t = __shfl_down(s, off, 64);
    if (lane + off < 64) s += t;
  }
  const T above = s - lsum;  // bins owned by lanes > l
  if (s >= target && (lane == 63 || above < target)) {
    // descend within this lane's 4 bins, highest first
    T cum = above;
    int bstar = lane * 4;
    T rem = target - cum;
#pragma unroll
    for (int j = 3; j >= 0; --j) {
      const int b = lane * 4 + j;
      const T c = vals[b];
      if (cum + c >= target) {
        bstar = b;
        rem = target - cum;
        break;
      }
      cum += c;
    }
    *s_b = (uint32_t)bstar;
    *s_rem = rem;
  }
  // degenerate: target exceeds the total (float rounding at p ~ 1.0) —
  // fall through to bin 0 like the serial walk
  if (lane == 0 && s < target) {
    *s_b = 0;
    *s_rem = target - s;
  }
}

template <int BLOCK>
__global__ __launch_bounds__(BLOCK) void sample_bf16_kernel(
    int32_t* __restrict__ out,             // [B]
    const uint16_t* __restrict__ logits,   // [B, V] bf16 bits
    const float* __restrict__ temperature, // [B]
    const int32_t* __restrict__ top_k,     // [B] (<=0 => disabled)
    const float* __restrict__ top_p,       // [B] (>=1 => disabled)
    const float* __restrict__ uniforms,    // [B]
    int V) {
  constexpr int NW = BLOCK / WAVE_SIZE;
  constexpr int REP = 2;  // bin replicas per wave (LDS-atomic conflict
                          // width vs footprint). Measured: REP=2 (~36 KB,
                          // 4 blocks/CU) == REP=4 (~68 KB, 2 blocks/CU)
                          // at B=512 — the histogram passes are
                          // per-element VALU-bound, not LDS- or
                          // occupancy-bound; keep the smaller footprint
  __shared__ float redf[NW];
  __shared__ uint32_t ubins[NW][REP][256];
  __shared__ float fbins[NW][REP][256];
  __shared__ uint32_t ucomb[256];
  __shared__ float fcomb[256];
  __shared__ uint32_t s_u32[4];
  __shared__ float s_f32[4];
  __shared__ float seg_sums[BLOCK + 1];

  const int seq = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const uint16_t* row = logits + (int64_t)seq * V;
  const float temp = temperature[seq];
  const bool greedy = (temp <= 0.0f);
  const float inv_t = greedy ? 1.0f : 1.0f / temp;
  const int V8 = V >> 3;
  const uint4* row8 = reinterpret_cast<const uint4*>(row);

  // ---- pass 1: unscaled max (order == scaled order) ----
  float mx = -INFINITY;
  for (int i = tid; i < V8; i += BLOCK) {
    const uint4 v = row8[i];
#pragma unroll
    for (int h = 0; h < 4; ++h) {
      const uint32_t q = (&v.x)[h];
      mx = fmaxf(mx, bf16_val((uint16_t)(q & 0xffffu)));
      mx = fmaxf(mx, bf16_val((uint16_t)(q >> 16)));
    }
  }
  for (int i = (V8 << 3) + tid; i < V; i += BLOCK)
    mx = fmaxf(mx, bf16_val(row[i]));
  mx = wave_reduce_max(mx);
  if (lane == 0) redf[wave] = mx;
  __syncthreads();
  if (tid == 0) {
    float r = redf[0];
    for (int w = 1; w < NW; ++w) r = fmaxf(r, redf[w]);
    s_f32[0] = r;
  }
  __syncthreads();
  const float M = s_f32[0];

  if (greedy) {
    if (tid == 0) s_u32[0] = 0xffffffffu;
    __syncthreads();
    for (int i = tid; i < V; i += BLOCK) {
      if (bf16_val(row[i]) == M) atomicMin(&s_u32[0], (uint32_t)i);
    }
    __syncthreads();
    if (tid == 0) out[seq] = (int32_t)s_u32[0];
    return;
  }

  const int k = top_k[seq];
  const float p = top_p[seq];
  const bool do_k = (k > 0 && k < V);
  const bool do_p = (p < 1.0f);
  uint32_t tau = 0;

  if (do_k || do_p) {
    uint32_t tk_prefix = 0, tp_prefix = 0;
    uint32_t tk_remaining = do_k ? (uint32_t)k : 0;
    float tp_target = 0.0f;
    const int rep = (lane >> 3) & (REP - 1);

    for (int level = 1; level >= 0; --level) {
      for (int b = tid; b < 256; b += BLOCK) {
#pragma unroll
        for (int w = 0; w < NW; ++w)
#pragma unroll
          for (int r = 0; r < REP; ++r) {
            ubins[w][r][b] = 0;
            fbins[w][r][b] = 0.0f;
          }
      }
      __syncthreads();
      const int shift = level * 8;
      uint32_t* const my_u = ubins[wave][rep];
      float* const my_f = fbins[wave][rep];
      // run-merge accumulators live across the WHOLE pass: near-flat
      // logits keep hitting the same bin, so each thread issues only a
      // handful of LDS atomics instead of one per 8 elements
      uint32_t runb_k = 0xffffffffu, runc_k = 0;
      uint32_t runb_p = 0xffffffffu;
      float runw_p = 0.0f;
      for (int i = tid; i < V8; i += BLOCK) {
        const uint4 v = row8[i];
#pragma unroll
        for (int h = 0; h < 8; ++h) {
          const uint32_t q = (&v.x)[h >> 1];
          const uint16_t u = (uint16_t)((h & 1) ? (q >> 16) : (q & 0xffffu));
          const uint32_t key = bf16_key(u);
          const uint32_t hi = (level == 1) ? 0 : (key >> 8);
          const uint32_t b = (key >> shift) & 0xff;
          if (do_k && (level == 1 || hi == tk_prefix)) {
            if (b == runb_k) {
              ++runc_k;
            } else {
              if (runc_k) atomicAdd(&my_u[runb_k], runc_k);
              runb_k = b;
              runc_k = 1;
            }
          }
          if (level == 1 || hi == tp_prefix) {
            const float w = __expf((bf16_val(u) - M) * inv_t);
            if (b == runb_p) {
              runw_p += w;
            } else {
              if (runb_p != 0xffffffffu) atomicAdd(&my_f[runb_p], runw_p);
              runb_p = b;
              runw_p = w;
            }
          }
        }
      }
      if (runc_k) atomicAdd(&my_u[runb_k], runc_k);
      if (runb_p != 0xffffffffu) atomicAdd(&my_f[runb_p], runw_p);
      for (int i = (V8 << 3) + tid; i < V; i += BLOCK) {
        const uint16_t u = row[i];
        const uint32_t key = bf16_key(u);
        const uint32_t hi = (level == 1) ? 0 : (key >> 8);
        const uint32_t b = (key >> shift) & 0xff;
        if (do_k && (level == 1 || hi == tk_prefix))
          atomicAdd(&my_u[b], 1u);
        if (level == 1 || hi == tp_prefix)
          atomicAdd(&my_f[b], __expf((bf16_val(u) - M) * inv_t));
      }
      __syncthreads();
      // combine replicas (parallel: one thread per bin)
      for (int b = tid; b < 256; b += BLOCK) {
        uint32_t cu = 0;
        float cf = 0.0f;
#pragma unroll
        for (int w = 0; w < NW; ++w)
#pragma unroll
          for (int r = 0; r < REP; ++r) {
            cu += ubins[w][r][b];
            cf += fbins[w][r][b];
          }
        ucomb[b] = cu;
        fcomb[b] = cf;
      }
      __syncthreads();
      if (level == 1) {
        // Z = sum of the weight bins (parallel block reduce)
        float z = (tid < 256) ? fcomb[tid] : 0.0f;
        z = wave_reduce_sum(z);
        if (lane == 0) redf[wave] = z;
        __syncthreads();
        if (tid == 0) {
          float zz = 0.0f;
          for (int w = 0; w < NW; ++w) zz += redf[w];
          s_f32[1] = zz;
        }
        __syncthreads();
        if (do_p) tp_target = p * s_f32[1];
      }
      // wave 0: top-k count search; wave 1: top-p mass search
      if (do_k && wave == 0) {
        suffix_threshold_search<uint32_t>(ucomb, tk_remaining, lane,
                                          &s_u32[0], &s_u32[1]);
      }
      if (do_p && wave == (NW > 1 ? 1 : 0)) {
        suffix_threshold_search<float>(fcomb, tp_target, lane, &s_u32[2],
                                       &s_f32[2]);
      }
      __syncthreads();
      if (do_k) {
        tk_prefix = (tk_prefix << 8) | (s_u32[0] & 0xffu);
        tk_remaining = s_u32[1];
      }
      if (do_p) {
        tp_prefix = (tp_prefix << 8) | (s_u32[2] & 0xffu);
        tp_target = s_f32[2];
      }
      __syncthreads();
    }
    tau = max(do_k ? tk_prefix : 0u, do_p ? tp_prefix : 0u);
  }

  // ---- kept mass + per-thread segment sums (one fused pass,
  //      vectorized uint4 loads; segments 8-aligned) ----
  const int seg = (((V + BLOCK - 1) / BLOCK) + 7) & ~7;
  const int s0 = min(V, tid * seg);
  const int s1 = min(V, s0 + seg);
  float segsum = 0.0f;
  int i = s0;
  for (; i + 8 <= s1; i += 8) {
    const uint4 v = *reinterpret_cast<const uint4*>(row + i);
#pragma unroll
    for (int h = 0; h < 8; ++h) {
      const uint32_t q = (&v.x)[h >> 1];
      const uint16_t u = (uint16_t)((h & 1) ? (q >> 16) : (q & 0xffffu));
      if (bf16_key(u) >= tau)
        segsum += __expf((bf16_val(u) - M) * inv_t);
    }
  }
  for (; i < s1; ++i) {
    const uint16_t u = row[i];
    if (bf16_key(u) >= tau) segsum += __expf((bf16_val(u) - M) * inv_t);
  }
  seg_sums[tid] = segsum;
  __syncthreads();
  if (tid == 0) {
    float run = 0.0f;
    for (int t = 0; t < BLOCK; ++t) {
      const float s = seg_sums[t];
      seg_sums[t] = run;
      run += s;
    }
    seg_sums[BLOCK] = run;  // = Zf
    s_u32[0] = 0xffffffffu;
  }
  __syncthreads();

  // ---- inverse-CDF walk in index order ----
  const float zf = seg_sums[BLOCK];
  const float target = uniforms[seq] * zf;
  const float base = seg_sums[tid];
  const float next = seg_sums[tid + 1];
  if (target >= base && (target < next || tid == BLOCK - 1)) {
    float run = base;
    int chosen = -1;
    for (int i = s0; i < s1; ++i) {
      const uint16_t u = row[i];
      if (bf16_key(u) >= tau) {
        run += __expf((bf16_val(u) - M) * inv_t);
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
      // numeric corner (u ~ 1.0): fall back to argmax
      c = 0;
      float best = -INFINITY;
      for (int i = 0; i < V; ++i) {
        const float x = bf16_val(row[i]);
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
  TORCH_CHECK(logits.is_cuda());
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(out.dtype() == torch::kInt32);
  const int B = logits.size(0);
  const int V = logits.size(1);
  constexpr int BLOCK = 256;
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  if (logits.dtype() == torch::kBFloat16) {
    // 8 waves per row: the radix walk is latency-bound at 4 waves
    // (PMC: 50% parked / 32% issue-stalled at BLOCK=256)
    constexpr int BLOCK_BF16 = 512;
    hipLaunchKernelGGL((sample_bf16_kernel<BLOCK_BF16>), dim3(B),
                       dim3(BLOCK_BF16), 0,
                       stream, (int32_t*)out.data_ptr(),
                       (const uint16_t*)logits.data_ptr(),
                       (const float*)temperature.data_ptr(),
                       (const int32_t*)top_k.data_ptr(),
                       (const float*)top_p.data_ptr(),
                       (const float*)uniforms.data_ptr(), V);
    HIP_CHECK_KERNEL();
    return;
  }
  TORCH_CHECK(logits.dtype() == torch::kFloat32);
  hipLaunchKernelGGL((sample_kernel<BLOCK>), dim3(B), dim3(BLOCK), 0, stream,
                     (int32_t*)out.data_ptr(), (const float*)logits.data_ptr(),
                     (const float*)temperature.data_ptr(),
                     (const int32_t*)top_k.data_ptr(),
                     (const float*)top_p.data_ptr(),
                     (const float*)uniforms.data_ptr(), V);
  HIP_CHECK_KERNEL();
}

}  // namespace igw
