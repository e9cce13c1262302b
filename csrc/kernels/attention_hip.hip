#include "hip/hip_runtime.h"
// Fused causal attention (flash-style) on CDNA4 MFMA matrix cores.
//
// Replaces the reference's materialized (B,H,T,T) score path
// (/root/reference/example/model.py:29-51) with an MI355X-native design:
//   forward : per Q block (8 waves x 16 rows), stream 64-key K/V tiles
//             through XOR-swizzled LDS; S^T = mfma(K, Q) per 16x16 tile
//             (v_mfma_f32_16x16x32_bf16, fp32 accum) so the softmax row is
//             lane-local; online softmax in exp2 space (log2e folded into
//             the Q prescale); P repacked to MFMA A-fragments with
//             tile-uniform ds_bpermute shuffles; O = P V with V staged
//             transposed. Saves per-row logsumexp (natural log).
//   backward: recompute-based two-kernel scheme (no atomics):
//             dkv kernel owns a 128-key block and accumulates dK/dV over Q
//             tiles; dq kernel owns a 128-row Q block. delta = rowsum(dO*O)
//             by a small wave-reduction kernel. The 1/sqrt(D) factor on
//             dS is folded into the dK/dQ epilogue.
//
// VALU-discipline (first profile showed 22:1 VALU:MFMA): every LDS offset
// (fragment reads, staging stores) is computed once before the K/V loop;
// causal masking runs only on diagonal tiles (wave-uniform branch); waves
// whose rows lie entirely above/below a tile skip its compute.
//
// Fragment layouts (gfx950, verified on hardware by scripts/debug_mfma.py
// and against ck_tile/ops/gemm/warp/warp_gemm_attribute_mfma_impl.hpp):
//   A[16x32] : lane l, elem e(0..7) -> A[l%16][(l/16)*8 + e]
//   B[32x16] : lane l, elem e      -> B[(l/16)*8 + e][l%16]
//   C[16x16] : lane l, reg  r(0..3)-> C[(l/16)*4 + r][l%16]
// (The hardware pairs A/B by register slot, so A and B must agree on the
// slot->k map; the C map is fixed.)
//
// Contract: bf16 tensors (B,H,T,64) contiguous, T % 64 == 0. The Python op
// (ops/attention.py) falls back to the composite path otherwise.
#include "common.h"

#define MFMA_BF16_16x16x32(a, b, c) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

namespace {

constexpr int KVB = 64;   // kv keys per LDS tile
constexpr int D = 64;     // head dim (all GPT-2 sizes)
constexpr float LOG2E = 1.4426950408889634f;
constexpr float LN2 = 0.6931471805599453f;

typedef __attribute__((ext_vector_type(8))) short bfrag;

// XOR swizzle for a [64][64] bf16 LDS image with 128-byte rows (Guideline 4).
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

// Per-tensor global strides in ELEMENTS (last dim must be contiguous).
// Lets the kernels consume the packed qkv projection / (B,T,H,D) activation
// layouts directly — no transpose-copies on the hot path.
struct GStride {
  long long b;  // batch stride
  long long h;  // head stride
  int t;        // row (token) stride
};

// Per-lane precomputed addressing for one [64][64] staging + fragment-read
// pattern. Staging is 512 chunks of 8 bf16 over NT threads.
struct TileAddr {
  int stage_src[2];   // global element offsets of this thread's chunks
  int stage_dst[2];   // swizzled LDS byte offsets (row-major image)
};

template <int NT>
DEV_INLINE TileAddr tile_addr(int tid, int t_stride) {
  TileAddr a;
#pragma unroll
  for (int rep = 0; rep < 512 / (NT / 8) / 8; ++rep) {
    int chunk = tid + rep * NT;
    a.stage_src[rep] = (chunk >> 3) * t_stride + (chunk & 7) * 8;
    a.stage_dst[rep] = swz(chunk >> 3, (chunk & 7) * 16);
  }
  return a;
}

template <int NT>
DEV_INLINE void stage_rowmajor(const bf16* __restrict__ g, char* lds,
                               const TileAddr& a) {
#pragma unroll
  for (int rep = 0; rep < 512 * 8 / NT / 8; ++rep)
    *reinterpret_cast<short8v*>(lds + a.stage_dst[rep]) =
        load8(g + a.stage_src[rep]);
}

// Transposed staging: dest row = source col. Scalar u16 writes.
template <int NT>
DEV_INLINE void stage_transposed(const bf16* __restrict__ g, char* lds,
                                 const TileAddr& a, const int (&tdst)[2][8]) {
#pragma unroll
  for (int rep = 0; rep < 512 * 8 / NT / 8; ++rep) {
    short8v v = load8(g + a.stage_src[rep]);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      *reinterpret_cast<short*>(lds + tdst[rep][j]) = v[j];
  }
}

template <int NT>
DEV_INLINE void transposed_dst(int tid, int (&tdst)[2][8]) {
#pragma unroll
  for (int rep = 0; rep < 512 * 8 / NT / 8; ++rep) {
    int chunk = tid + rep * NT;
    int row = chunk >> 3;
    int c0 = (chunk & 7) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) tdst[rep][j] = swz(c0 + j, row * 2);
  }
}

// C-layout -> A-fragment redistribution (see header comment of cshuffle).
struct CShuffled {
  unsigned sh[4][2][2];  // [tile][word][half]
};

DEV_INLINE CShuffled cshuffle(const unsigned pw[4][2], int g, int c) {
  const int L0 = 32 * (g & 1) + c;
  CShuffled s;
#pragma unroll
  for (int qt = 0; qt < 4; ++qt)
#pragma unroll
    for (int wd = 0; wd < 2; ++wd) {
      s.sh[qt][wd][0] = __shfl(pw[qt][wd], L0, WAVE);
      s.sh[qt][wd][1] = __shfl(pw[qt][wd], L0 + 16, WAVE);
    }
  return s;
}

template <int KS>
DEV_INLINE bfrag frag_from_shuffled(const CShuffled& s, int g) {
  const bool hi = (g >> 1) != 0;
  union { bfrag f; unsigned w[4]; } r;
  r.w[0] = hi ? s.sh[2 * KS + 1][0][0] : s.sh[2 * KS][0][0];
  r.w[1] = hi ? s.sh[2 * KS + 1][1][0] : s.sh[2 * KS][1][0];
  r.w[2] = hi ? s.sh[2 * KS + 1][0][1] : s.sh[2 * KS][0][1];
  r.w[3] = hi ? s.sh[2 * KS + 1][1][1] : s.sh[2 * KS][1][1];
  return r.f;
}

// load a global B/A-style fragment (row r0+c, 8 elems at col c0+8g) scaled
DEV_INLINE bfrag load_frag_scaled(const bf16* p, int row, int col, int st,
                                  float s) {
  short8v v = load8(p + (long long)row * st + col);
  union { bfrag f; short w[8]; } r;
#pragma unroll
  for (int e = 0; e < 8; ++e) r.w[e] = bf_pack(bf_elem(v, e) * s);
  return r.f;
}

DEV_INLINE bfrag load_frag(const bf16* p, int row, int col, int st) {
  union { bfrag f; short8v v; } r;
  r.v = load8(p + (long long)row * st + col);
  return r.f;
}

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------
template <int NW>
__launch_bounds__(NW * WAVE, 2)
__global__ void attn_fwd_kernel(const bf16* __restrict__ q,
                                const bf16* __restrict__ k,
                                const bf16* __restrict__ v,
                                bf16* __restrict__ o, float* __restrict__ lse,
                                int T, int H, float scale, GStride sq,
                                GStride so) {
  constexpr int BM = NW * 16;  // q rows per workgroup
  constexpr int NT = NW * WAVE;
  __shared__ __attribute__((aligned(16))) char smem[2 * KVB * D * 2];
  char* lds_k = smem;                 // [64][64] keys row-major
  char* lds_vt = smem + KVB * D * 2;  // [64(d)][64(key)] V transposed

  const int qb = blockIdx.x;
  const long long bh = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid / WAVE;
  const int g = lane >> 4;
  const int c = lane & 15;

  const long long boff = (bh / H) * sq.b + (bh % H) * sq.h;
  const bf16* qp = q + boff + (long long)(qb * BM) * sq.t;
  const bf16* kp = k + boff;
  const bf16* vp = v + boff;

  // addressing, hoisted out of the K/V loop
  const TileAddr ta = tile_addr<NT>(tid, sq.t);
  int tdst[2][8];
  transposed_dst<NT>(tid, tdst);
  int kf_off[4][2], vf_off[4][2];
#pragma unroll
  for (int t16 = 0; t16 < 4; ++t16)
#pragma unroll
    for (int ds = 0; ds < 2; ++ds) {
      kf_off[t16][ds] = swz(t16 * 16 + c, (ds * 32 + 8 * g) * 2);
      vf_off[t16][ds] = kf_off[t16][ds];  // same pattern on the V^T image
    }

  // Q fragments, prescaled by scale*log2e (softmax runs in exp2 space)
  const float qs = scale * LOG2E;
  bfrag q_frag[2];
#pragma unroll
  for (int ds = 0; ds < 2; ++ds)
    q_frag[ds] = load_frag_scaled(qp, w * 16 + c, ds * 32 + 8 * g, sq.t, qs);

  f32x4 o_acc[4] = {};
  float m_run = -INFINITY;
  float l_run = 0.f;

  const int row_lo = qb * BM + w * 16;       // this wave's first q row
  const int row_me = row_lo + c;             // this lane's q row
  const int n_kv = (qb + 1) * BM / KVB;
  for (int j = 0; j < n_kv; ++j) {
    __syncthreads();
    stage_rowmajor<NT>(kp + (long long)(j * KVB) * sq.t, lds_k, ta);
    stage_transposed<NT>(vp + (long long)(j * KVB) * sq.t, lds_vt, ta, tdst);
    __syncthreads();

    const int key0 = j * KVB;
    if (key0 > row_lo + 15) continue;        // wave fully above this tile

    // S^T tiles: C[key = 16*sub + 4g + r][qrow = c]  (values are log2-scaled)
    f32x4 st[4];
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      f32x4 acc = {};
      acc = MFMA_BF16_16x16x32(lds_read16(lds_k, kf_off[sub][0]), q_frag[0], acc);
      acc = MFMA_BF16_16x16x32(lds_read16(lds_k, kf_off[sub][1]), q_frag[1], acc);
      st[sub] = acc;
    }

    const bool diag = key0 + KVB - 1 > row_lo;  // some key may exceed a row
    float mt = -INFINITY;
    if (diag) {
#pragma unroll
      for (int sub = 0; sub < 4; ++sub)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = key0 + sub * 16 + 4 * g + r;
          float s = (key <= row_me) ? st[sub][r] : -INFINITY;
          st[sub][r] = s;
          mt = fmaxf(mt, s);
        }
    } else {
#pragma unroll
      for (int sub = 0; sub < 4; ++sub)
#pragma unroll
        for (int r = 0; r < 4; ++r) mt = fmaxf(mt, st[sub][r]);
    }
    mt = fmaxf(mt, __shfl_xor(mt, 16, WAVE));
    mt = fmaxf(mt, __shfl_xor(mt, 32, WAVE));
    const float m_new = fmaxf(m_run, mt);
    const float alpha = exp2f(m_run - m_new);
    float psum = 0.f;
    unsigned pw[4][2];
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      // masked scores are -inf and m_new is finite, so exp2 gives exact 0
      const float p0 = exp2f(st[sub][0] - m_new);
      const float p1 = exp2f(st[sub][1] - m_new);
      const float p2 = exp2f(st[sub][2] - m_new);
      const float p3 = exp2f(st[sub][3] - m_new);
      psum += p0 + p1 + p2 + p3;
      pw[sub][0] = pack2(p0, p1);
      pw[sub][1] = pack2(p2, p3);
    }
    psum += __shfl_xor(psum, 16, WAVE);
    psum += __shfl_xor(psum, 32, WAVE);
    l_run = l_run * alpha + psum;
    m_run = m_new;

    // O rescale (skip when alpha == 1 for every row of the wave)
    if (__any(alpha != 1.0f)) {
      float alpha_row[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) alpha_row[r] = __shfl(alpha, 4 * g + r, WAVE);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt)
#pragma unroll
        for (int r = 0; r < 4; ++r) o_acc[dt][r] *= alpha_row[r];
    }

    // PV
    const CShuffled psh = cshuffle(pw, g, c);
    const bfrag pa0 = frag_from_shuffled<0>(psh, g);
    const bfrag pa1 = frag_from_shuffled<1>(psh, g);
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      o_acc[dt] = MFMA_BF16_16x16x32(pa0, lds_read16(lds_vt, vf_off[dt][0]),
                                     o_acc[dt]);
      o_acc[dt] = MFMA_BF16_16x16x32(pa1, lds_read16(lds_vt, vf_off[dt][1]),
                                     o_acc[dt]);
    }
  }

  // epilogue: O /= l ; lse = (m + log2(l)) * ln2
  float linv_row[4];
#pragma unroll
  for (int r = 0; r < 4; ++r)
    linv_row[r] = 1.0f / __shfl(l_run, 4 * g + r, WAVE);
  bf16* op = o + (bh / H) * so.b + (bh % H) * so.h
             + (long long)(qb * BM + w * 16) * so.t;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      op[(4 * g + r) * so.t + dt * 16 + c] = f2bf(o_acc[dt][r] * linv_row[r]);
  if (lane < 16) {
    lse[bh * T + qb * BM + w * 16 + c] = (m_run + log2f(l_run)) * LN2;
  }
}

// ---------------------------------------------------------------------------
// Backward: delta = rowsum(dO * O)
// ---------------------------------------------------------------------------
__global__ void attn_delta_kernel(const bf16* __restrict__ dout,
                                  const bf16* __restrict__ o,
                                  float* __restrict__ delta, long long R,
                                  int T, int H, GStride so) {
  const long long row = (long long)blockIdx.x * (blockDim.x / WAVE)
                        + threadIdx.x / WAVE;
  if (row >= R) return;
  const int lane = threadIdx.x & (WAVE - 1);
  const long long bh = row / T;
  const long long off = (bh / H) * so.b + (bh % H) * so.h
                        + (long long)(row % T) * so.t + lane;
  float acc = bf2f(dout[off]) * bf2f(o[off]);
  acc = wave_sum(acc);
  if (lane == 0) delta[row] = acc;
}

// ---------------------------------------------------------------------------
// Backward dK/dV: one workgroup per (NW*16)-key block; wave w owns 16 keys.
// ---------------------------------------------------------------------------
template <int NW>
__launch_bounds__(NW * WAVE, 2)
__global__ void attn_bwd_dkv_kernel(const bf16* __restrict__ q,
                                    const bf16* __restrict__ k,
                                    const bf16* __restrict__ v,
                                    const bf16* __restrict__ dout,
                                    const float* __restrict__ lse,
                                    const float* __restrict__ delta,
                                    bf16* __restrict__ dk, bf16* __restrict__ dv,
                                    int T, int H, float scale, GStride sq,
                                    GStride so, GStride sd) {
  constexpr int BK = NW * 16;  // keys per workgroup
  constexpr int NT = NW * WAVE;
  __shared__ __attribute__((aligned(16))) char smem[4 * KVB * D * 2 + 2 * KVB * 4];
  char* lds_q = smem;
  char* lds_do = smem + KVB * D * 2;
  char* lds_qt = smem + 2 * KVB * D * 2;
  char* lds_dot = smem + 3 * KVB * D * 2;
  float* lds_lse = reinterpret_cast<float*>(smem + 4 * KVB * D * 2);
  float* lds_dlt = lds_lse + KVB;

  const int jb = blockIdx.x;
  const long long bh = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid / WAVE;
  const int g = lane >> 4;
  const int c = lane & 15;

  const long long boff = (bh / H) * sq.b + (bh % H) * sq.h;
  const long long ooff = (bh / H) * so.b + (bh % H) * so.h;
  const bf16* qp = q + boff;
  const bf16* kp = k + boff;
  const bf16* vp = v + boff;
  const bf16* dop = dout + ooff;

  const TileAddr ta = tile_addr<NT>(tid, sq.t);
  const TileAddr tao = tile_addr<NT>(tid, so.t);
  int tdst[2][8];
  transposed_dst<NT>(tid, tdst);
  int af_off[4][2], bf_off[4][2];
#pragma unroll
  for (int t16 = 0; t16 < 4; ++t16)
#pragma unroll
    for (int ds = 0; ds < 2; ++ds) {
      af_off[t16][ds] = swz(t16 * 16 + c, (ds * 32 + 8 * g) * 2);
      bf_off[t16][ds] = af_off[t16][ds];
    }

  // This wave's K (prescaled by scale*log2e) and V fragments in registers.
  const float ks_scale = scale * LOG2E;
  const int key_lo = jb * BK + w * 16;   // first key of this wave
  const int key_me = key_lo + c;         // this lane's key
  bfrag k_frag[2], v_frag[2];
#pragma unroll
  for (int ds = 0; ds < 2; ++ds) {
    k_frag[ds] = load_frag_scaled(kp, jb * BK + w * 16 + c, ds * 32 + 8 * g,
                                  sq.t, ks_scale);
    v_frag[ds] = load_frag(vp, jb * BK + w * 16 + c, ds * 32 + 8 * g, sq.t);
  }

  f32x4 dk_acc[4] = {};
  f32x4 dv_acc[4] = {};

  for (int i = jb * BK / KVB; i < T / KVB; ++i) {
    __syncthreads();
    stage_rowmajor<NT>(qp + (long long)(i * KVB) * sq.t, lds_q, ta);
    stage_rowmajor<NT>(dop + (long long)(i * KVB) * so.t, lds_do, tao);
    stage_transposed<NT>(qp + (long long)(i * KVB) * sq.t, lds_qt, ta, tdst);
    stage_transposed<NT>(dop + (long long)(i * KVB) * so.t, lds_dot, tao, tdst);
    if (tid < KVB) {
      lds_lse[tid] = lse[bh * T + i * KVB + tid] * LOG2E;
      lds_dlt[tid] = delta[bh * T + i * KVB + tid];
    }
    __syncthreads();

    const int q0 = i * KVB;
    if (q0 + KVB - 1 < key_lo) continue;   // all rows above this wave's keys

    const bool diag = q0 < key_lo + 16;    // some row may precede a key
    unsigned pwp[4][2];
    unsigned pws[4][2];
#pragma unroll
    for (int qt = 0; qt < 4; ++qt) {
      f32x4 s_acc = {};
      f32x4 dp_acc = {};
      s_acc = MFMA_BF16_16x16x32(lds_read16(lds_q, af_off[qt][0]), k_frag[0], s_acc);
      s_acc = MFMA_BF16_16x16x32(lds_read16(lds_q, af_off[qt][1]), k_frag[1], s_acc);
      dp_acc = MFMA_BF16_16x16x32(lds_read16(lds_do, af_off[qt][0]), v_frag[0], dp_acc);
      dp_acc = MFMA_BF16_16x16x32(lds_read16(lds_do, af_off[qt][1]), v_frag[1], dp_acc);
      float p[4], dsv[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + qt * 16 + 4 * g + r;
        const float l2 = lds_lse[qt * 16 + 4 * g + r];
        const float dlt = lds_dlt[qt * 16 + 4 * g + r];
        float pp = exp2f(s_acc[r] - l2);
        if (diag) pp = (key_me <= qrow) ? pp : 0.f;
        p[r] = pp;
        dsv[r] = pp * (dp_acc[r] - dlt);   // scale folded into epilogue
      }
      pwp[qt][0] = pack2(p[0], p[1]);
      pwp[qt][1] = pack2(p[2], p[3]);
      pws[qt][0] = pack2(dsv[0], dsv[1]);
      pws[qt][1] = pack2(dsv[2], dsv[3]);
    }

    const CShuffled pshp = cshuffle(pwp, g, c);
    const CShuffled pshs = cshuffle(pws, g, c);
    const bfrag ap0 = frag_from_shuffled<0>(pshp, g);
    const bfrag ap1 = frag_from_shuffled<1>(pshp, g);
    const bfrag as0 = frag_from_shuffled<0>(pshs, g);
    const bfrag as1 = frag_from_shuffled<1>(pshs, g);
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      dv_acc[dt] = MFMA_BF16_16x16x32(ap0, lds_read16(lds_dot, bf_off[dt][0]), dv_acc[dt]);
      dv_acc[dt] = MFMA_BF16_16x16x32(ap1, lds_read16(lds_dot, bf_off[dt][1]), dv_acc[dt]);
      dk_acc[dt] = MFMA_BF16_16x16x32(as0, lds_read16(lds_qt, bf_off[dt][0]), dk_acc[dt]);
      dk_acc[dt] = MFMA_BF16_16x16x32(as1, lds_read16(lds_qt, bf_off[dt][1]), dk_acc[dt]);
    }
  }

  const long long doff = (bh / H) * sd.b + (bh % H) * sd.h
                         + (long long)(jb * BK + w * 16) * sd.t;
  bf16* dkp = dk + doff;
  bf16* dvp = dv + doff;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      dkp[(4 * g + r) * sd.t + dt * 16 + c] = f2bf(dk_acc[dt][r] * scale);
      dvp[(4 * g + r) * sd.t + dt * 16 + c] = f2bf(dv_acc[dt][r]);
    }
}

// ---------------------------------------------------------------------------
// Backward dQ: one workgroup per (NW*16)-row Q block; wave w owns 16 rows.
// ---------------------------------------------------------------------------
template <int NW>
__launch_bounds__(NW * WAVE, 2)
__global__ void attn_bwd_dq_kernel(const bf16* __restrict__ q,
                                   const bf16* __restrict__ k,
                                   const bf16* __restrict__ v,
                                   const bf16* __restrict__ dout,
                                   const float* __restrict__ lse,
                                   const float* __restrict__ delta,
                                   bf16* __restrict__ dq, int T, int H,
                                   float scale, GStride sq, GStride so,
                                   GStride sd) {
  constexpr int BM = NW * 16;
  constexpr int NT = NW * WAVE;
  __shared__ __attribute__((aligned(16))) char smem[3 * KVB * D * 2];
  char* lds_k = smem;                     // K row-major (A of S^T)
  char* lds_kt = smem + KVB * D * 2;      // K^T (B of dQ)
  char* lds_v = smem + 2 * KVB * D * 2;   // V row-major (A of dP^T)

  const int qb = blockIdx.x;
  const long long bh = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid / WAVE;
  const int g = lane >> 4;
  const int c = lane & 15;

  const long long boff = (bh / H) * sq.b + (bh % H) * sq.h;
  const long long ooff = (bh / H) * so.b + (bh % H) * so.h;
  const bf16* qp = q + boff + (long long)(qb * BM) * sq.t;
  const bf16* kp = k + boff;
  const bf16* vp = v + boff;
  const bf16* dop = dout + ooff + (long long)(qb * BM) * so.t;

  const TileAddr ta = tile_addr<NT>(tid, sq.t);
  int tdst[2][8];
  transposed_dst<NT>(tid, tdst);
  int f_off[4][2];
#pragma unroll
  for (int t16 = 0; t16 < 4; ++t16)
#pragma unroll
    for (int ds = 0; ds < 2; ++ds)
      f_off[t16][ds] = swz(t16 * 16 + c, (ds * 32 + 8 * g) * 2);

  const float qs = scale * LOG2E;
  bfrag q_frag[2], do_frag[2];
#pragma unroll
  for (int ds = 0; ds < 2; ++ds) {
    q_frag[ds] = load_frag_scaled(qp, w * 16 + c, ds * 32 + 8 * g, sq.t, qs);
    do_frag[ds] = load_frag(dop, w * 16 + c, ds * 32 + 8 * g, so.t);
  }
  const int row_lo = qb * BM + w * 16;
  const int row_me = row_lo + c;
  const float lse2_me = lse[bh * T + row_me] * LOG2E;
  const float dlt_me = delta[bh * T + row_me];

  f32x4 dq_acc[4] = {};

  const int n_kv = (qb + 1) * BM / KVB;
  for (int j = 0; j < n_kv; ++j) {
    __syncthreads();
    stage_rowmajor<NT>(kp + (long long)(j * KVB) * sq.t, lds_k, ta);
    stage_transposed<NT>(kp + (long long)(j * KVB) * sq.t, lds_kt, ta, tdst);
    stage_rowmajor<NT>(vp + (long long)(j * KVB) * sq.t, lds_v, ta);
    __syncthreads();

    const int key0 = j * KVB;
    if (key0 > row_lo + 15) continue;

    const bool diag = key0 + KVB - 1 > row_lo;
    unsigned pws[4][2];
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      f32x4 s_acc = {};
      f32x4 dp_acc = {};
      s_acc = MFMA_BF16_16x16x32(lds_read16(lds_k, f_off[sub][0]), q_frag[0], s_acc);
      s_acc = MFMA_BF16_16x16x32(lds_read16(lds_k, f_off[sub][1]), q_frag[1], s_acc);
      dp_acc = MFMA_BF16_16x16x32(lds_read16(lds_v, f_off[sub][0]), do_frag[0], dp_acc);
      dp_acc = MFMA_BF16_16x16x32(lds_read16(lds_v, f_off[sub][1]), do_frag[1], dp_acc);
      float dsv[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int key = key0 + sub * 16 + 4 * g + r;
        float pp = exp2f(s_acc[r] - lse2_me);
        if (diag) pp = (key <= row_me) ? pp : 0.f;
        dsv[r] = pp * (dp_acc[r] - dlt_me);  // scale folded into epilogue
      }
      pws[sub][0] = pack2(dsv[0], dsv[1]);
      pws[sub][1] = pack2(dsv[2], dsv[3]);
    }

    const CShuffled pshs = cshuffle(pws, g, c);
    const bfrag as0 = frag_from_shuffled<0>(pshs, g);
    const bfrag as1 = frag_from_shuffled<1>(pshs, g);
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      dq_acc[dt] = MFMA_BF16_16x16x32(as0, lds_read16(lds_kt, f_off[dt][0]), dq_acc[dt]);
      dq_acc[dt] = MFMA_BF16_16x16x32(as1, lds_read16(lds_kt, f_off[dt][1]), dq_acc[dt]);
    }
  }

  bf16* dqp = dq + (bh / H) * sd.b + (bh % H) * sd.h
              + (long long)(qb * BM + w * 16) * sd.t;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      dqp[(4 * g + r) * sd.t + dt * 16 + c] = f2bf(dq_acc[dt][r] * scale);
}

}  // namespace

extern "C" {

// strides arrays: {b, h, t} in elements, per tensor group:
// sq = q/k/v, so = o (and dO in bwd), sd = dq/dk/dv.
hipError_t tdsa_attn_fwd(const void* q, const void* k, const void* v, void* o,
                         float* lse, long long B, long long H, int T,
                         float scale, const long long* sq_in,
                         const long long* so_in, hipStream_t stream) {
  if (T % KVB) return hipErrorInvalidValue;
  GStride sq{sq_in[0], sq_in[1], (int)sq_in[2]};
  GStride so{so_in[0], so_in[1], (int)so_in[2]};
  if (T % 128 == 0) {
    dim3 grid(T / 128, B * H);
    hipLaunchKernelGGL(attn_fwd_kernel<8>, grid, dim3(512), 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v, (bf16*)o,
                       lse, T, (int)H, scale, sq, so);
  } else {
    dim3 grid(T / 64, B * H);
    hipLaunchKernelGGL(attn_fwd_kernel<4>, grid, dim3(256), 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v, (bf16*)o,
                       lse, T, (int)H, scale, sq, so);
  }
  return hipGetLastError();
}

hipError_t tdsa_attn_bwd(const void* q, const void* k, const void* v,
                         const void* o, const float* lse, const void* dout,
                         void* dq, void* dk, void* dv, float* delta,
                         long long B, long long H, int T, float scale,
                         const long long* sq_in, const long long* so_in,
                         const long long* sd_in, hipStream_t stream) {
  if (T % KVB) return hipErrorInvalidValue;
  GStride sq{sq_in[0], sq_in[1], (int)sq_in[2]};
  GStride so{so_in[0], so_in[1], (int)so_in[2]};
  GStride sd{sd_in[0], sd_in[1], (int)sd_in[2]};
  const long long BH = B * H;
  const long long R = BH * T;
  {
    const int rows_per_block = 256 / WAVE;
    const long long grid = (R + rows_per_block - 1) / rows_per_block;
    hipLaunchKernelGGL(attn_delta_kernel, dim3(grid), dim3(256), 0, stream,
                       (const bf16*)dout, (const bf16*)o, delta, R, T, (int)H,
                       so);
  }
  if (T % 128 == 0) {
    dim3 grid(T / 128, BH);
    hipLaunchKernelGGL(attn_bwd_dkv_kernel<8>, grid, dim3(512), 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v,
                       (const bf16*)dout, lse, delta, (bf16*)dk, (bf16*)dv, T,
                       (int)H, scale, sq, so, sd);
    hipLaunchKernelGGL(attn_bwd_dq_kernel<8>, grid, dim3(512), 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v,
                       (const bf16*)dout, lse, delta, (bf16*)dq, T, (int)H,
                       scale, sq, so, sd);
  } else {
    dim3 grid(T / 64, BH);
    hipLaunchKernelGGL(attn_bwd_dkv_kernel<4>, grid, dim3(256), 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v,
                       (const bf16*)dout, lse, delta, (bf16*)dk, (bf16*)dv, T,
                       (int)H, scale, sq, so, sd);
    hipLaunchKernelGGL(attn_bwd_dq_kernel<4>, grid, dim3(256), 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v,
                       (const bf16*)dout, lse, delta, (bf16*)dq, T, (int)H,
                       scale, sq, so, sd);
  }
  return hipGetLastError();
}

}  // extern "C"
