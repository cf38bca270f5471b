// Shared helpers for sentio_amd gfx950 (CDNA4) kernels.
// Pure HIP — no torch headers here (fast TU compiles); bindings.hip owns torch.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64  // CDNA wavefront is 64 lanes (cdna_hip_programming.md §1)

typedef __hip_bfloat16 bf16;

// vector types for wide loads (guide G13: always vectorize bf16 loads)
typedef short  short8 __attribute__((ext_vector_type(8)));
typedef short  short4v __attribute__((ext_vector_type(4)));
typedef float  f32x4 __attribute__((ext_vector_type(4)));
typedef float  f32x16 __attribute__((ext_vector_type(16)));
typedef short  bf16x8 __attribute__((ext_vector_type(8)));
typedef short  bf16x4 __attribute__((ext_vector_type(4)));

// Non-temporal 16 B load (MI355X_MICROARCH.md nt-weights: streamed data one
// CU reads exactly once should bypass cache retention — measured -18% landing
// time on weight streams; use for decode KV streaming, NOT for reused tiles).
__device__ __forceinline__ bf16x8 nt_load8(const short* p) {
  return __builtin_nontemporal_load(reinterpret_cast<const bf16x8*>(p));
}

__device__ __forceinline__ float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2bf(float v) { return __float2bfloat16(v); }

__device__ __forceinline__ float bits2f(short s) {
  union { unsigned u; float f; } c;
  c.u = ((unsigned)(unsigned short)s) << 16;
  return c.f;
}
__device__ __forceinline__ short f2bits(float f) {
  union { unsigned u; float f; } c;
  c.f = f;
  // round-to-nearest-even bf16
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;
  return (short)(c.u >> 16);
}

// wave-wide reductions (64 lanes)
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}
__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}
// reduction within 16-lane groups (MFMA C-frag rows)
__device__ __forceinline__ float group16_max(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}
__device__ __forceinline__ float group16_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// block reduction via LDS (expects <= 1024 threads; scratch is 32 floats)
__device__ __forceinline__ float block_sum(float v, float* scratch) {
  int wid = threadIdx.x / WAVE;
  int nw = blockDim.x / WAVE;
  v = wave_sum(v);
  if ((threadIdx.x & (WAVE - 1)) == 0) scratch[wid] = v;
  __syncthreads();
  float out = (threadIdx.x < nw) ? scratch[threadIdx.x] : 0.f;
  if (threadIdx.x < WAVE) out = wave_sum(out);
  if (threadIdx.x == 0) scratch[0] = out;
  __syncthreads();
  out = scratch[0];
  __syncthreads();
  return out;
}
__device__ __forceinline__ float block_max(float v, float* scratch) {
  int wid = threadIdx.x / WAVE;
  int nw = blockDim.x / WAVE;
  v = wave_max(v);
  if ((threadIdx.x & (WAVE - 1)) == 0) scratch[wid] = v;
  __syncthreads();
  float out = (threadIdx.x < nw) ? scratch[threadIdx.x] : -INFINITY;
  if (threadIdx.x < WAVE) out = wave_max(out);
  if (threadIdx.x == 0) scratch[0] = out;
  __syncthreads();
  out = scratch[0];
  __syncthreads();
  return out;
}

// Batched block reduction: G values per thread reduced across the block in
// ONE barrier pair (vs 3 barriers × G calls of block_max/block_sum).
// scratch must hold G * (blockDim/WAVE) floats.
template <int G, bool IS_MAX>
__device__ __forceinline__ void block_reduce_vec(float (&v)[G],
                                                 float* scratch) {
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
#pragma unroll
  for (int g = 0; g < G; ++g) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float o = __shfl_xor(v[g], off, WAVE);
      v[g] = IS_MAX ? fmaxf(v[g], o) : v[g] + o;
    }
    if (lane == 0) scratch[g * nw + wid] = v[g];
  }
  __syncthreads();
#pragma unroll
  for (int g = 0; g < G; ++g) {
    float acc = IS_MAX ? -INFINITY : 0.f;
    for (int w = 0; w < nw; ++w) {
      float x = scratch[g * nw + w];
      acc = IS_MAX ? fmaxf(acc, x) : acc + x;
    }
    v[g] = acc;
  }
  __syncthreads();  // scratch reusable after return
}

// simple splitmix-style hash for on-device RNG (sampling kernel)
__device__ __forceinline__ unsigned hash_u32(unsigned a, unsigned b, unsigned c) {
  unsigned h = a * 0x9E3779B9u ^ b * 0x85EBCA6Bu ^ c * 0xC2B2AE35u;
  h ^= h >> 16; h *= 0x7FEB352Du;
  h ^= h >> 15; h *= 0x846CA68Bu;
  h ^= h >> 16;
  return h;
}

#define HIP_CHECK_LAUNCH()                                                    \
  do {                                                                        \
    hipError_t e = hipGetLastError();                                         \
    if (e != hipSuccess) return e;                                            \
  } while (0)
