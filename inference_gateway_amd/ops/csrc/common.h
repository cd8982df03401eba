// Common helpers for MI355X (gfx950 / CDNA4) HIP kernels.
//
// Design notes: CDNA4 wavefront is 64 lanes; block dims are multiples of 64.
// All kernels here are written directly for gfx950 — no CUDA compatibility
// paths, no Triton. bf16 is the primary activation dtype.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define WAVE_SIZE 64

#define DEVINLINE __device__ __forceinline__

// ---------------------------------------------------------------------------
// bf16 <-> f32
// ---------------------------------------------------------------------------
DEVINLINE float bf16_to_f32(uint16_t h) {
  union {
    uint32_t u;
    float f;
  } v;
  v.u = ((uint32_t)h) << 16;
  return v.f;
}

DEVINLINE uint16_t f32_to_bf16(float f) {
  union {
    float f;
    uint32_t u;
  } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t lsb = (v.u >> 16) & 1;
  v.u += 0x7fffu + lsb;
  return (uint16_t)(v.u >> 16);
}

// packed vector types for wide loads
typedef uint32_t u32;
typedef uint16_t u16;
struct alignas(16) ushort8 {
  uint16_t v[8];
};
struct alignas(8) ushort4x {
  uint16_t v[4];
};

// ---------------------------------------------------------------------------
// Wave-level reductions (64-wide).
// ---------------------------------------------------------------------------
DEVINLINE float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    x += __shfl_xor(x, off, WAVE_SIZE);
  }
  return x;
}

DEVINLINE float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    x = fmaxf(x, __shfl_xor(x, off, WAVE_SIZE));
  }
  return x;
}

// Reduce over a contiguous group of `width` lanes (width must be power of 2).
template <int WIDTH>
DEVINLINE float group_reduce_sum(float x) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) {
    x += __shfl_xor(x, off, WAVE_SIZE);
  }
  return x;
}

template <int WIDTH>
DEVINLINE float group_reduce_max(float x) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) {
    x = fmaxf(x, __shfl_xor(x, off, WAVE_SIZE));
  }
  return x;
}

// Block-level reduce using LDS scratch; BLOCK must be multiple of 64.
// `lds` must have at least BLOCK/64 floats. Result valid on all threads.
template <int BLOCK>
DEVINLINE float block_reduce_sum(float x, float* lds) {
  const int nwaves = BLOCK / WAVE_SIZE;
  int lane = threadIdx.x & (WAVE_SIZE - 1);
  int wave = threadIdx.x >> 6;
  x = wave_reduce_sum(x);
  if (lane == 0) lds[wave] = x;
  __syncthreads();
  float r = (threadIdx.x < nwaves) ? lds[threadIdx.x] : 0.0f;
#pragma unroll
  for (int off = 1; off < nwaves; off <<= 1) {
    r += __shfl_xor(r, off, WAVE_SIZE);
  }
  // broadcast from lane 0 of wave 0 via lds
  if (threadIdx.x == 0) lds[0] = r;
  __syncthreads();
  return lds[0];
}

#define HIP_CHECK_KERNEL()                                                     \
  do {                                                                         \
    hipError_t err = hipGetLastError();                                        \
    if (err != hipSuccess) {                                                   \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(err)); \
    }                                                                          \
  } while (0)
