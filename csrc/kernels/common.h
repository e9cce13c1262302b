// Common device helpers for the tiny_deepspeed_amd CDNA4 (gfx950) kernels.
//
// MI355X facts used throughout (see /opt/skills/guides/MI355X_MICROARCH.md):
//   wavefront = 64 lanes; 256 CUs in 8 XCDs; 160 KiB LDS/CU; HBM3E ~8 TB/s.
// All kernels accumulate in fp32 (acc-dtype policy of SURVEY.md 2.10D) and
// vectorize bf16 global traffic as short4/short8 reinterprets (Guideline 13).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

typedef __hip_bfloat16 bf16;

// ---- vector types ---------------------------------------------------------
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

DEV_INLINE float bf2f(bf16 x) { return __bfloat162float(x); }
DEV_INLINE bf16 f2bf(float x) { return __float2bfloat16(x); }

// Load/store 8 bf16 (16 B) per lane — the coalescing sweet spot.
DEV_INLINE short8v load8(const bf16* p) {
  return *reinterpret_cast<const short8v*>(p);
}
DEV_INLINE void store8(bf16* p, short8v v) {
  *reinterpret_cast<short8v*>(p) = v;
}
DEV_INLINE float bf_elem(short8v v, int i) {
  union { short s; bf16 b; } u; u.s = v[i]; return bf2f(u.b);
}
DEV_INLINE short bf_pack(float x) {
  union { short s; bf16 b; } u; u.b = f2bf(x); return u.s;
}

// ---- wave reductions ------------------------------------------------------
DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}
DEV_INLINE float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block-level reductions through LDS (blockDim.x threads, <=1024).
// `scratch` must hold blockDim.x/WAVE floats.
DEV_INLINE float block_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  v = wave_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (lane < nw) ? scratch[lane] : 0.0f;
  r = wave_sum(r);
  // every wave now computed the same value in lane set; broadcast via lane 0
  r = __shfl(r, 0, WAVE);
  __syncthreads();
  return r;
}
// Paired reduction: one barrier round for two sums (layernorm's sum/sumsq
// and c1/c2 pairs — halves the barrier count of two block_sum calls).
DEV_INLINE void block_sum2(float& a, float& b, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  a = wave_sum(a);
  b = wave_sum(b);
  if (lane == 0) {
    scratch[wid] = a;
    scratch[nw + wid] = b;
  }
  __syncthreads();
  float ra = (lane < nw) ? scratch[lane] : 0.0f;
  float rb = (lane < nw) ? scratch[nw + lane] : 0.0f;
  ra = wave_sum(ra);
  rb = wave_sum(rb);
  a = __shfl(ra, 0, WAVE);
  b = __shfl(rb, 0, WAVE);
  __syncthreads();
}

DEV_INLINE float block_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  v = wave_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (lane < nw) ? scratch[lane] : -INFINITY;
  r = wave_max(r);
  r = __shfl(r, 0, WAVE);
  __syncthreads();
  return r;
}

#define HIP_CHECK(cmd)                                                     \
  do {                                                                     \
    hipError_t e_ = (cmd);                                                 \
    if (e_ != hipSuccess) {                                                \
      return e_;                                                           \
    }                                                                      \
  } while (0)

// Grid sizing for memory-bound grid-stride kernels (Guideline 11):
// cap at ~8 blocks per CU and stride the rest.
DEV_INLINE int global_tid() { return blockIdx.x * blockDim.x + threadIdx.x; }
DEV_INLINE int global_nthreads() { return gridDim.x * blockDim.x; }

inline int ln_grid(long long work, int block) {
  long long g = (work + block - 1) / block;
  const long long cap = 2048;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return static_cast<int>(g);
}
