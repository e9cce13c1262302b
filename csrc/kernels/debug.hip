// MFMA layout probes (debug/selftest). These validate the fragment-layout
// assumptions the attention kernels rely on, directly on hardware:
//   dbg_mfma  : one v_mfma_f32_16x16x32_bf16 from global A[16][32], B[32][16]
//               under a selectable A/B fragment-layout variant; D written via
//               the assumed C layout.
//   dbg_stage : stage a [64][64] bf16 tile through swizzled LDS (row-major or
//               transposed) and reconstruct it through frag_read — validates
//               the staging/swizzle/fragment-read path without MFMA.
#include "common.h"

namespace tdsa {

typedef __attribute__((ext_vector_type(8))) short bfrag;

DEV_INLINE int swz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

__global__ void dbg_mfma_kernel(const bf16* __restrict__ A,
                                const bf16* __restrict__ B,
                                float* __restrict__ D, int variant) {
  const int lane = threadIdx.x;
  const int g = lane >> 4;
  const int c = lane & 15;
  union { bfrag f; short s[8]; } a, b;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    int k;
    if (variant == 0) k = 8 * g + e;                        // contiguous
    else k = (e < 4 ? 0 : 16) + 4 * g + (e & 3);            // split-K16
    a.s[e] = *reinterpret_cast<const short*>(A + c * 32 + k);
    b.s[e] = *reinterpret_cast<const short*>(B + k * 16 + c);
  }
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.f, b.f, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[(4 * g + r) * 16 + c] = acc[r];
}

// 32x32x16 probe with the PRODUCTION layouts (attention.hip):
// A[l%32][(l/32)*8+e], B[(l/32)*8+e][l%32], C[(r&3)+8*(r>>2)+4*(l/32)][l%32]
__global__ void dbg_mfma32_kernel(const bf16* __restrict__ A,
                                  const bf16* __restrict__ B,
                                  float* __restrict__ D) {
  const int lane = threadIdx.x;
  const int m32 = lane & 31;
  const int h = lane >> 5;
  union { bfrag f; short s[8]; } a, b;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int k = 8 * h + e;
    a.s[e] = *reinterpret_cast<const short*>(A + m32 * 16 + k);
    b.s[e] = *reinterpret_cast<const short*>(B + k * 32 + m32);
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a.f, b.f, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    D[((r & 3) + 8 * (r >> 2) + 4 * h) * 32 + m32] = acc[r];
}

__global__ void dbg_stage_kernel(const bf16* __restrict__ in,
                                 bf16* __restrict__ out, int transposed) {
  __shared__ __attribute__((aligned(16))) char lds[64 * 64 * 2];
  // stage (same code as attention.hip)
#pragma unroll
  for (int rep = 0; rep < 2; ++rep) {
    int chunk = threadIdx.x + rep * 256;
    int row = chunk >> 3;
    int c0 = (chunk & 7) * 8;
    short8v v = load8(in + chunk * 8);
    if (!transposed) {
      *reinterpret_cast<short8v*>(lds + swz(row, c0 * 2)) = v;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *reinterpret_cast<short*>(lds + swz(c0 + j, row * 2)) = v[j];
    }
  }
  __syncthreads();
  // read back through the vector-read addressing (for the transposed image
  // this validates scalar swizzled writes against 16B swizzled reads)
#pragma unroll
  for (int rep = 0; rep < 2; ++rep) {
    int chunk = threadIdx.x + rep * 256;
    int row = chunk >> 3;
    int c0 = (chunk & 7) * 8;
    short8v v = *reinterpret_cast<const short8v*>(lds + swz(row, c0 * 2));
    store8(out + chunk * 8, v);
  }
}

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4v;
#define LDS_AS __attribute__((address_space(3)))

// Probe ds_read_tr16_b64 lane mapping: stage a [64][64] bf16 image (swizzled
// row-major), then issue tr reads with the attention kernels' intended
// addressing and dump every (dt, s, half, j, lane) result.
__global__ void dbg_tr16_kernel(const bf16* __restrict__ in,
                                float* __restrict__ out) {
  __shared__ __attribute__((aligned(16))) char lds[64 * 64 * 2];
#pragma unroll
  for (int rep = 0; rep < 8; ++rep) {  // 64 threads stage 512 chunks
    int chunk = threadIdx.x + rep * 64;
    int row = chunk >> 3;
    int c0 = (chunk & 7) * 8;
    *reinterpret_cast<short8v*>(lds + swz(row, c0 * 2)) = load8(in + chunk * 8);
  }
  __syncthreads();
  const int l = threadIdx.x;
  const int g4 = l >> 4;
  const int i = l & 15;
  for (int dt = 0; dt < 2; ++dt)
    for (int s = 0; s < 4; ++s)
      for (int half = 0; half < 2; ++half) {
        // intended block: rows 16s + 8*(g4>>1) + 4*half + (i/4),
        //                 cols dt*32 + 16*(g4&1) + 4*(i%4)
        const int row = 16 * s + 8 * (g4 >> 1) + 4 * half + (i >> 2);
        const int colbyte = (dt * 32 + 16 * (g4 & 1) + 4 * (i & 3)) * 2;
        const int addr = swz(row, colbyte);
        auto p = (LDS_AS bf16x4v*)(lds + addr);
        bf16x4v r = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          out[(((dt * 4 + s) * 2 + half) * 4 + j) * 64 + l] = (float)r[j];
      }
}

}  // namespace tdsa

using namespace tdsa;

extern "C" {

hipError_t tdsa_dbg_tr16(const void* in, float* out, hipStream_t stream) {
  hipLaunchKernelGGL(dbg_tr16_kernel, dim3(1), dim3(64), 0, stream,
                     (const bf16*)in, out);
  return hipGetLastError();
}

hipError_t tdsa_dbg_mfma32(const void* A, const void* B, float* D,
                           hipStream_t stream) {
  hipLaunchKernelGGL(dbg_mfma32_kernel, dim3(1), dim3(64), 0, stream,
                     (const bf16*)A, (const bf16*)B, D);
  return hipGetLastError();
}

hipError_t tdsa_dbg_mfma(const void* A, const void* B, float* D, int variant,
                         hipStream_t stream) {
  hipLaunchKernelGGL(dbg_mfma_kernel, dim3(1), dim3(64), 0, stream,
                     (const bf16*)A, (const bf16*)B, D, variant);
  return hipGetLastError();
}

hipError_t tdsa_dbg_stage(const void* in, void* out, int transposed,
                          hipStream_t stream) {
  hipLaunchKernelGGL(dbg_stage_kernel, dim3(1), dim3(256), 0, stream,
                     (const bf16*)in, (bf16*)out, transposed);
  return hipGetLastError();
}

}  // extern "C"
