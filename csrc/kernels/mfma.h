// Shared CDNA4 MFMA fragment / LDS-staging machinery (gfx950).
//
// Extracted from the attention kernel so the hand-written GEMMs reuse the
// hardware-verified pieces (layout probes: scripts/debug_mfma.py and
// tests/test_gpu_kernels.py::test_mfma_layout_probes_gpu):
//
// v_mfma_f32_32x32x16_bf16 fragment layouts (slot-pairing rule):
//   A[32x16]: lane l, elem e(0..7) -> A[l%32][(l/32)*8 + e]
//   B[16x32]: lane l, elem e      -> B[(l/32)*8 + e][l%32]
//   C[32x32]: lane l, reg r(0..15)-> C[(r&3) + 8*(r>>2) + 4*(l/32)][l%32]
//
// Note the A/B symmetry: both map lane->(k-slice element, 32-axis index)
// identically, so BOTH operands of a GEMM whose reduction dim is the ROW
// dim of a row-major [64][64] LDS image come from the same
// ds_read_tr16_b64 hardware-transpose read (tr_bfrag below).
#pragma once

#include "common.h"

#define MFMA32(a, b, c) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16((a), (b), (c), 0, 0, 0)

namespace tdsa {

typedef __attribute__((ext_vector_type(8))) short bfrag;
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4v;
#define LDS_AS __attribute__((address_space(3)))

// XOR swizzle for a [64][64] bf16 LDS image with 128-byte rows.
DEV_INLINE int swz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

DEV_INLINE unsigned pack2(float lo, float hi) {
  union { struct { short a, b; } s; unsigned u; } u;
  u.s.a = bf_pack(lo);
  u.s.b = bf_pack(hi);
  return u.u;
}

DEV_INLINE bfrag lds_read16(const char* lds, int byte_off) {
  return *reinterpret_cast<const bfrag*>(lds + byte_off);
}

// Row index of C register r for this half-wave (the 32x32 C layout).
DEV_INLINE int crow(int r, int h32) { return (r & 3) + 8 * (r >> 2) + 4 * h32; }

// Per-lane staging state for [64][64] bf16 tiles: absolute per-thread source
// pointers (bumped by a constant per tile) + swizzled LDS byte offsets.
// 512 chunks of 8 bf16 over NT threads.
template <int NT>
struct Stage {
  static constexpr int REPS = 512 / NT;
  const bf16* src[REPS];
  int dst[REPS];
  long long step;  // elements to advance per tile

  DEV_INLINE Stage() {}

  DEV_INLINE Stage(const bf16* g, int tid, long long t_stride, int rows = 64) {
    init(g, tid, t_stride, rows);
  }

  DEV_INLINE void init(const bf16* g, int tid, long long t_stride,
                       int rows = 64) {
    step = (long long)rows * t_stride;
#pragma unroll
    for (int rep = 0; rep < REPS; ++rep) {
      int chunk = tid + rep * NT;
      src[rep] = g + (long long)(chunk >> 3) * t_stride + (chunk & 7) * 8;
      dst[rep] = swz(chunk >> 3, (chunk & 7) * 16);
    }
  }

  DEV_INLINE void run(char* lds) {
#pragma unroll
    for (int rep = 0; rep < REPS; ++rep)
      *reinterpret_cast<short8v*>(lds + dst[rep]) = load8(src[rep]);
  }

  DEV_INLINE void advance() {
#pragma unroll
    for (int rep = 0; rep < REPS; ++rep) src[rep] += step;
  }

  // split fetch/put: issue the global loads early (they fly under the
  // previous tile's compute), write to LDS later.
  DEV_INLINE void fetch(short8v (&buf)[REPS]) {
#pragma unroll
    for (int rep = 0; rep < REPS; ++rep) buf[rep] = load8(src[rep]);
  }

  DEV_INLINE void put(char* lds, const short8v (&buf)[REPS]) {
#pragma unroll
    for (int rep = 0; rep < REPS; ++rep)
      *reinterpret_cast<short8v*>(lds + dst[rep]) = buf[rep];
  }
};

// MFMA operand via gfx950 hardware transpose read (ds_read_tr16_b64): two
// reads deliver k = 16*S + 8*(l/32) + e over ROWS of a row-major swizzled
// [64][64] image at this lane's column — no transposed LDS image needed.
// Lane mapping hardware-verified by scripts/debug_mfma.py::probe_tr16.
DEV_INLINE bfrag tr_bfrag(const char* lds, int a_lo, int a_hi) {
  bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (LDS_AS bf16x4v*)(lds + a_lo));
  bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (LDS_AS bf16x4v*)(lds + a_hi));
  union { bfrag f; bf16x4v h[2]; } r;
  r.h[0] = lo;
  r.h[1] = hi;
  return r.f;
}

// Per-lane tr-read base addresses: trb[dt][half] for the 32-col tile dt;
// slice S adds S*2048 bytes (16 rows). Lane i=l&15 supplies the chunk at
// row (i>>2) of the 4-row block, byte quarter 4*(i&3).
DEV_INLINE void tr_bases(int lane, int (&trb)[2][2]) {
  const int g4 = lane >> 4;
  const int i = lane & 15;
  const int hh = g4 >> 1;
  const int iq = i >> 2;
#pragma unroll
  for (int dt = 0; dt < 2; ++dt)
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int row = 8 * hh + 4 * half + iq;
      const int colbyte = (dt * 32 + 16 * (g4 & 1) + 4 * (i & 3)) * 2;
      trb[dt][half] = row * 128 + (colbyte ^ ((row & 7) << 4));
    }
}

}  // namespace tdsa
