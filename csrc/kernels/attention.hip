// Fused causal attention (flash-style) on CDNA4 MFMA matrix cores.
//
// Replaces the reference's materialized (B,H,T,T) score path
// (/root/reference/example/model.py:29-51) with an MI355X-native design:
//   forward : per 64-row Q block, stream 64-key K/V tiles through
//             XOR-swizzled LDS; S^T = mfma(K, Q) per 16x16 tile
//             (v_mfma_f32_16x16x32_bf16, fp32 accum) so the softmax row is
//             lane-local; online softmax; P repacked to MFMA A-fragments
//             with 8 ds_bpermute shuffles (no LDS round trip); O = P V with
//             V staged transposed. Saves per-row logsumexp.
//   backward: recompute-based two-kernel scheme (no atomics):
//             dkv kernel owns a 64-key block and accumulates dK/dV over Q
//             tiles; dq kernel owns a 64-row Q block. delta = rowsum(dO*O)
//             precomputed by a small wave-reduction kernel.
//
// Fragment layouts (gfx950, verified against rocm CK headers
// ck_tile/ops/gemm/warp/warp_gemm_attribute_mfma_impl.hpp:195-221):
//   A[16x32] : lane l, elem e(0..7) -> A[l%16][(l/16)*8 + e]
//   B[32x16] : lane l, elem e      -> B[(l/16)*8 + e][l%16]
//   C[16x16] : lane l, reg  r(0..3)-> C[(l/16)*4 + r][l%16]
//
// Contract: bf16 tensors (B,H,T,64) contiguous, T % 64 == 0. The Python op
// (ops/attention.py) falls back to the composite path otherwise.
#include "common.h"

#define MFMA_BF16_16x16x32(a, b, c) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

namespace {

constexpr int BLK = 64;   // q-rows / kv-keys per workgroup tile
constexpr int D = 64;     // head dim (all GPT-2 sizes)
constexpr int NW = 4;     // waves per workgroup

typedef __attribute__((ext_vector_type(8))) short bfrag;  // 8 bf16 (4 VGPRs)

// XOR swizzle for a [64][64] bf16 LDS image with 128-byte rows: spreads the
// 16-lane b128 read groups over 8 slots (Guideline 4: <=2-way).
DEV_INLINE int swz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

// Stage a [64][64] bf16 tile row-major from global into LDS, swizzled.
// 256 threads, fully coalesced global reads (16B per lane, linear).
DEV_INLINE void stage_rowmajor(const bf16* __restrict__ g, char* lds) {
#pragma unroll
  for (int rep = 0; rep < 2; ++rep) {
    int chunk = threadIdx.x + rep * 256;  // 512 chunks of 8 bf16
    int row = chunk >> 3;
    int byte = (chunk & 7) * 16;
    short8v v = load8(g + chunk * 8);
    *reinterpret_cast<short8v*>(lds + swz(row, byte)) = v;
  }
}

// Stage transposed: LDS image [d_or_col][64] from a global [64][d] tile.
// Scalar u16 LDS writes (one-time per tile, shared by 4 waves).
DEV_INLINE void stage_transposed(const bf16* __restrict__ g, char* lds) {
#pragma unroll
  for (int rep = 0; rep < 2; ++rep) {
    int chunk = threadIdx.x + rep * 256;
    int row = chunk >> 3;            // source row (key/qrow)
    int c0 = (chunk & 7) * 8;        // source col (d)
    short8v v = load8(g + chunk * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int trow = c0 + j;             // dest row = d
      *reinterpret_cast<short*>(lds + swz(trow, row * 2)) = v[j];
    }
  }
}

// Read one A/B fragment (16B) from a swizzled [64][64] image:
// lane (g=l/16, c=l%16) reads row `row16 + c`, 8 bf16 at col `col8 + 8*g`.
DEV_INLINE bfrag frag_read(const char* lds, int row16, int col8, int g, int c) {
  int row = row16 + c;
  return *reinterpret_cast<const bfrag*>(lds + swz(row, (col8 + 8 * g) * 2));
}

DEV_INLINE unsigned pack2(float lo, float hi) {
  union { struct { short a, b; } s; unsigned u; } u;
  u.s.a = bf_pack(lo);
  u.s.b = bf_pack(hi);
  return u.u;
}

// Redistribute per-lane C-layout values (packed bf16 words w0=(r0,r1),
// w1=(r2,r3) per 16-wide tile) into MFMA A-fragments whose k axis runs over
// the C tiles' row axis. Dest lane (g,c), slice ks, elem e needs the value
// of tile qt = 2*ks + (g>>1) from source lane 32*(g&1) + (e>=4 ? 16 : 0) + c.
//
// __shfl evaluates its operand on the SOURCE lane, so the tile index inside
// the shuffled expression must not depend on the destination's registers:
// shuffle every tile's words once (uniform expressions), then select.
struct CShuffled {
  unsigned sh[4][2][2];  // [tile][word][half: src L0 / L0+16]
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
  const bool hi = (g >> 1) != 0;  // tile = 2*KS + (g>>1): select, no scratch
  union { bfrag f; unsigned w[4]; } r;
  r.w[0] = hi ? s.sh[2 * KS + 1][0][0] : s.sh[2 * KS][0][0];
  r.w[1] = hi ? s.sh[2 * KS + 1][1][0] : s.sh[2 * KS][1][0];
  r.w[2] = hi ? s.sh[2 * KS + 1][0][1] : s.sh[2 * KS][0][1];
  r.w[3] = hi ? s.sh[2 * KS + 1][1][1] : s.sh[2 * KS][1][1];
  return r.f;
}

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------
__launch_bounds__(256, 2)
__global__ void attn_fwd_kernel(const bf16* __restrict__ q,
                                const bf16* __restrict__ k,
                                const bf16* __restrict__ v,
                                bf16* __restrict__ o, float* __restrict__ lse,
                                int T, float scale) {
  __shared__ __attribute__((aligned(16))) char smem[2 * BLK * D * 2];
  char* lds_k = smem;                 // [64][64] keys row-major
  char* lds_vt = smem + BLK * D * 2;  // [64(d)][64(key)] V transposed

  const int qb = blockIdx.x;          // q block
  const long long bh = blockIdx.y;    // batch*head
  const int lane = threadIdx.x & (WAVE - 1);
  const int w = threadIdx.x / WAVE;   // wave id: owns q rows w*16..w*16+15
  const int g = lane >> 4;
  const int c = lane & 15;

  const bf16* qp = q + (bh * T + qb * BLK) * D;
  const bf16* kp = k + bh * T * D;
  const bf16* vp = v + bh * T * D;

  // Q B-fragments for this wave (rows w*16+c, 2 d-slices), pre-scaled into
  // the softmax instead (scale applied to S).
  bfrag q_frag[2];
#pragma unroll
  for (int ds = 0; ds < 2; ++ds)
    q_frag[ds] = *reinterpret_cast<const bfrag*>(
        qp + (w * 16 + c) * D + ds * 32 + 8 * g);

  f32x4 o_acc[4] = {};
  float m_run = -INFINITY;
  float l_run = 0.f;

  const int n_kv = qb + 1;  // causal: kv tiles 0..qb
  for (int j = 0; j < n_kv; ++j) {
    __syncthreads();
    stage_rowmajor(kp + j * BLK * D, lds_k);
    stage_transposed(vp + j * BLK * D, lds_vt);
    __syncthreads();

    // S^T tiles: C[key = 16*sub + 4g + r][qrow = c]
    f32x4 st[4];
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      f32x4 acc = {};
#pragma unroll
      for (int ds = 0; ds < 2; ++ds) {
        bfrag a = frag_read(lds_k, sub * 16, ds * 32, g, c);
        acc = MFMA_BF16_16x16x32(a, q_frag[ds], acc);
      }
      st[sub] = acc;
    }

    // scale + causal mask + online softmax (row = this lane's qrow)
    const int qrow = qb * BLK + w * 16 + c;
    float mt = -INFINITY;
#pragma unroll
    for (int sub = 0; sub < 4; ++sub)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int key = j * BLK + sub * 16 + 4 * g + r;
        float s = st[sub][r] * scale;
        s = (key <= qrow) ? s : -INFINITY;
        st[sub][r] = s;
        mt = fmaxf(mt, s);
      }
    mt = fmaxf(mt, __shfl_xor(mt, 16, WAVE));
    mt = fmaxf(mt, __shfl_xor(mt, 32, WAVE));
    const float m_new = fmaxf(m_run, mt);
    const float alpha = __expf(m_run - m_new);  // m_run=-inf,m_new=-inf can't happen (key 0 valid)
    float psum = 0.f;
    unsigned pw[4][2];
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      float p0, p1, p2, p3;
      p0 = (st[sub][0] == -INFINITY) ? 0.f : __expf(st[sub][0] - m_new);
      p1 = (st[sub][1] == -INFINITY) ? 0.f : __expf(st[sub][1] - m_new);
      p2 = (st[sub][2] == -INFINITY) ? 0.f : __expf(st[sub][2] - m_new);
      p3 = (st[sub][3] == -INFINITY) ? 0.f : __expf(st[sub][3] - m_new);
      psum += p0 + p1 + p2 + p3;
      pw[sub][0] = pack2(p0, p1);
      pw[sub][1] = pack2(p2, p3);
    }
    psum += __shfl_xor(psum, 16, WAVE);
    psum += __shfl_xor(psum, 32, WAVE);
    l_run = l_run * alpha + psum;
    m_run = m_new;

    // O rescale: O C-layout rows are qrow = 4g+r -> fetch that row's alpha
    float alpha_row[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) alpha_row[r] = __shfl(alpha, 4 * g + r, WAVE);
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[dt][r] *= alpha_row[r];

    // PV: O[qrow][d] += P[qrow][key] V[key][d]
    const CShuffled psh = cshuffle(pw, g, c);
    {
      bfrag pa0 = frag_from_shuffled<0>(psh, g);
      bfrag pa1 = frag_from_shuffled<1>(psh, g);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        bfrag vb0 = frag_read(lds_vt, dt * 16, 0, g, c);
        bfrag vb1 = frag_read(lds_vt, dt * 16, 32, g, c);
        o_acc[dt] = MFMA_BF16_16x16x32(pa0, vb0, o_acc[dt]);
        o_acc[dt] = MFMA_BF16_16x16x32(pa1, vb1, o_acc[dt]);
      }
    }
  }

  // epilogue: O /= l ; lse = m + log(l)
  float linv_row[4];
#pragma unroll
  for (int r = 0; r < 4; ++r)
    linv_row[r] = 1.0f / __shfl(l_run, 4 * g + r, WAVE);
  bf16* op = o + (bh * T + qb * BLK + w * 16) * D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      op[(4 * g + r) * D + dt * 16 + c] = f2bf(o_acc[dt][r] * linv_row[r]);
  if (lane < 16) {
    lse[bh * T + qb * BLK + w * 16 + c] = m_run + __logf(l_run);
  }
}

// ---------------------------------------------------------------------------
// Backward: delta = rowsum(dO * O)
// ---------------------------------------------------------------------------
__global__ void attn_delta_kernel(const bf16* __restrict__ dout,
                                  const bf16* __restrict__ o,
                                  float* __restrict__ delta, long long R) {
  const long long row = (long long)blockIdx.x * (blockDim.x / WAVE)
                        + threadIdx.x / WAVE;
  if (row >= R) return;
  const int lane = threadIdx.x & (WAVE - 1);
  float acc = bf2f(dout[row * D + lane]) * bf2f(o[row * D + lane]);
  acc = wave_sum(acc);
  if (lane == 0) delta[row] = acc;
}

// ---------------------------------------------------------------------------
// Backward dK/dV: one workgroup per 64-key block; wave w owns keys w*16..+15.
// ---------------------------------------------------------------------------
__launch_bounds__(256, 2)
__global__ void attn_bwd_dkv_kernel(const bf16* __restrict__ q,
                                    const bf16* __restrict__ k,
                                    const bf16* __restrict__ v,
                                    const bf16* __restrict__ dout,
                                    const float* __restrict__ lse,
                                    const float* __restrict__ delta,
                                    bf16* __restrict__ dk, bf16* __restrict__ dv,
                                    int T, float scale) {
  // carves: Q row-major, dO row-major, Q^T, dO^T, lse+delta tiles
  __shared__ __attribute__((aligned(16))) char smem[4 * BLK * D * 2 + 2 * BLK * 4];
  char* lds_q = smem;
  char* lds_do = smem + BLK * D * 2;
  char* lds_qt = smem + 2 * BLK * D * 2;
  char* lds_dot = smem + 3 * BLK * D * 2;
  float* lds_lse = reinterpret_cast<float*>(smem + 4 * BLK * D * 2);
  float* lds_dlt = lds_lse + BLK;

  const int jb = blockIdx.x;        // kv block
  const long long bh = blockIdx.y;
  const int lane = threadIdx.x & (WAVE - 1);
  const int w = threadIdx.x / WAVE;  // wave owns keys w*16..w*16+15
  const int g = lane >> 4;
  const int c = lane & 15;

  const bf16* qp = q + bh * T * D;
  const bf16* kp = k + bh * T * D;
  const bf16* vp = v + bh * T * D;
  const bf16* dop = dout + bh * T * D;

  // This wave's K and V B-fragments (key = c, d = ds*32+8g+e) in registers.
  bfrag k_frag[2], v_frag[2];
#pragma unroll
  for (int ds = 0; ds < 2; ++ds) {
    k_frag[ds] = *reinterpret_cast<const bfrag*>(
        kp + (jb * BLK + w * 16 + c) * D + ds * 32 + 8 * g);
    v_frag[ds] = *reinterpret_cast<const bfrag*>(
        vp + (jb * BLK + w * 16 + c) * D + ds * 32 + 8 * g);
  }

  f32x4 dk_acc[4] = {};
  f32x4 dv_acc[4] = {};

  for (int i = jb; i < T / BLK; ++i) {
    __syncthreads();
    stage_rowmajor(qp + i * BLK * D, lds_q);
    stage_rowmajor(dop + i * BLK * D, lds_do);
    stage_transposed(qp + i * BLK * D, lds_qt);
    stage_transposed(dop + i * BLK * D, lds_dot);
    if (threadIdx.x < BLK) {
      lds_lse[threadIdx.x] = lse[bh * T + i * BLK + threadIdx.x];
      lds_dlt[threadIdx.x] = delta[bh * T + i * BLK + threadIdx.x];
    }
    __syncthreads();

    // S and dP tiles: C[qrow = 16*qt + 4g + r][key = w*16 + c]
    unsigned pwp[4][2];   // P packed
    unsigned pws[4][2];   // dS packed
#pragma unroll
    for (int qt = 0; qt < 4; ++qt) {
      f32x4 s_acc = {};
      f32x4 dp_acc = {};
#pragma unroll
      for (int ds = 0; ds < 2; ++ds) {
        bfrag aq = frag_read(lds_q, qt * 16, ds * 32, g, c);
        bfrag ado = frag_read(lds_do, qt * 16, ds * 32, g, c);
        s_acc = MFMA_BF16_16x16x32(aq, k_frag[ds], s_acc);
        dp_acc = MFMA_BF16_16x16x32(ado, v_frag[ds], dp_acc);
      }
      float p[4], dsv[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = i * BLK + qt * 16 + 4 * g + r;
        const int key = jb * BLK + w * 16 + c;
        const float l = lds_lse[qt * 16 + 4 * g + r];
        const float dlt = lds_dlt[qt * 16 + 4 * g + r];
        float pp = (key <= qrow) ? __expf(s_acc[r] * scale - l) : 0.f;
        p[r] = pp;
        dsv[r] = scale * pp * (dp_acc[r] - dlt);
      }
      pwp[qt][0] = pack2(p[0], p[1]);
      pwp[qt][1] = pack2(p[2], p[3]);
      pws[qt][0] = pack2(dsv[0], dsv[1]);
      pws[qt][1] = pack2(dsv[2], dsv[3]);
    }

    // dV[key][d] += P^T dO ; dK[key][d] += dS^T Q
    const CShuffled pshp = cshuffle(pwp, g, c);
    const CShuffled pshs = cshuffle(pws, g, c);
    {
      bfrag ap0 = frag_from_shuffled<0>(pshp, g);
      bfrag ap1 = frag_from_shuffled<1>(pshp, g);
      bfrag as0 = frag_from_shuffled<0>(pshs, g);
      bfrag as1 = frag_from_shuffled<1>(pshs, g);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        bfrag bdo0 = frag_read(lds_dot, dt * 16, 0, g, c);
        bfrag bdo1 = frag_read(lds_dot, dt * 16, 32, g, c);
        bfrag bq0 = frag_read(lds_qt, dt * 16, 0, g, c);
        bfrag bq1 = frag_read(lds_qt, dt * 16, 32, g, c);
        dv_acc[dt] = MFMA_BF16_16x16x32(ap0, bdo0, dv_acc[dt]);
        dv_acc[dt] = MFMA_BF16_16x16x32(ap1, bdo1, dv_acc[dt]);
        dk_acc[dt] = MFMA_BF16_16x16x32(as0, bq0, dk_acc[dt]);
        dk_acc[dt] = MFMA_BF16_16x16x32(as1, bq1, dk_acc[dt]);
      }
    }
  }

  // write: lane (g,c) holds rows key = w*16 + 4g + r, col = dt*16 + c
  bf16* dkp = dk + (bh * T + jb * BLK + w * 16) * D;
  bf16* dvp = dv + (bh * T + jb * BLK + w * 16) * D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      dkp[(4 * g + r) * D + dt * 16 + c] = f2bf(dk_acc[dt][r]);
      dvp[(4 * g + r) * D + dt * 16 + c] = f2bf(dv_acc[dt][r]);
    }
}

// ---------------------------------------------------------------------------
// Backward dQ: one workgroup per 64-row Q block; wave w owns rows w*16..+15.
// ---------------------------------------------------------------------------
__launch_bounds__(256, 2)
__global__ void attn_bwd_dq_kernel(const bf16* __restrict__ q,
                                   const bf16* __restrict__ k,
                                   const bf16* __restrict__ v,
                                   const bf16* __restrict__ dout,
                                   const float* __restrict__ lse,
                                   const float* __restrict__ delta,
                                   bf16* __restrict__ dq, int T, float scale) {
  __shared__ __attribute__((aligned(16))) char smem[3 * BLK * D * 2];
  char* lds_k = smem;                     // K row-major (A of S^T)
  char* lds_kt = smem + BLK * D * 2;      // K^T (B of dQ)
  char* lds_v = smem + 2 * BLK * D * 2;   // V row-major (A of dP^T)

  const int qb = blockIdx.x;
  const long long bh = blockIdx.y;
  const int lane = threadIdx.x & (WAVE - 1);
  const int w = threadIdx.x / WAVE;
  const int g = lane >> 4;
  const int c = lane & 15;

  const bf16* qp = q + (bh * T + qb * BLK) * D;
  const bf16* kp = k + bh * T * D;
  const bf16* vp = v + bh * T * D;
  const bf16* dop = dout + (bh * T + qb * BLK) * D;

  bfrag q_frag[2], do_frag[2];
#pragma unroll
  for (int ds = 0; ds < 2; ++ds) {
    q_frag[ds] = *reinterpret_cast<const bfrag*>(
        qp + (w * 16 + c) * D + ds * 32 + 8 * g);
    do_frag[ds] = *reinterpret_cast<const bfrag*>(
        dop + (w * 16 + c) * D + ds * 32 + 8 * g);
  }
  const int qrow_mine = qb * BLK + w * 16 + c;
  const float lse_mine = lse[bh * T + qrow_mine];
  const float dlt_mine = delta[bh * T + qrow_mine];

  f32x4 dq_acc[4] = {};

  for (int j = 0; j <= qb; ++j) {
    __syncthreads();
    stage_rowmajor(kp + j * BLK * D, lds_k);
    stage_transposed(kp + j * BLK * D, lds_kt);
    stage_rowmajor(vp + j * BLK * D, lds_v);
    __syncthreads();

    // S^T and dP^T tiles: C[key = 16*sub + 4g + r][qrow = c]
    unsigned pws[4][2];  // dS^T packed (k-index = key)
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      f32x4 s_acc = {};
      f32x4 dp_acc = {};
#pragma unroll
      for (int ds = 0; ds < 2; ++ds) {
        bfrag ak = frag_read(lds_k, sub * 16, ds * 32, g, c);
        bfrag av = frag_read(lds_v, sub * 16, ds * 32, g, c);
        s_acc = MFMA_BF16_16x16x32(ak, q_frag[ds], s_acc);
        dp_acc = MFMA_BF16_16x16x32(av, do_frag[ds], dp_acc);
      }
      float dsv[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int key = j * BLK + sub * 16 + 4 * g + r;
        float pp = (key <= qrow_mine)
                       ? __expf(s_acc[r] * scale - lse_mine) : 0.f;
        dsv[r] = scale * pp * (dp_acc[r] - dlt_mine);
      }
      pws[sub][0] = pack2(dsv[0], dsv[1]);
      pws[sub][1] = pack2(dsv[2], dsv[3]);
    }

    // dQ[qrow][d] += dS K
    const CShuffled pshs = cshuffle(pws, g, c);
    {
      bfrag as0 = frag_from_shuffled<0>(pshs, g);
      bfrag as1 = frag_from_shuffled<1>(pshs, g);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        bfrag bk0 = frag_read(lds_kt, dt * 16, 0, g, c);
        bfrag bk1 = frag_read(lds_kt, dt * 16, 32, g, c);
        dq_acc[dt] = MFMA_BF16_16x16x32(as0, bk0, dq_acc[dt]);
        dq_acc[dt] = MFMA_BF16_16x16x32(as1, bk1, dq_acc[dt]);
      }
    }
  }

  bf16* dqp = dq + (bh * T + qb * BLK + w * 16) * D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      dqp[(4 * g + r) * D + dt * 16 + c] = f2bf(dq_acc[dt][r]);
}

}  // namespace

extern "C" {

hipError_t tdsa_attn_fwd(const void* q, const void* k, const void* v, void* o,
                         float* lse, long long BH, int T, float scale,
                         hipStream_t stream) {
  if (T % BLK) return hipErrorInvalidValue;
  dim3 grid(T / BLK, BH);
  hipLaunchKernelGGL(attn_fwd_kernel, grid, dim3(256), 0, stream,
                     (const bf16*)q, (const bf16*)k, (const bf16*)v, (bf16*)o,
                     lse, T, scale);
  return hipGetLastError();
}

hipError_t tdsa_attn_bwd(const void* q, const void* k, const void* v,
                         const void* o, const float* lse, const void* dout,
                         void* dq, void* dk, void* dv, float* delta,
                         long long BH, int T, float scale, hipStream_t stream) {
  if (T % BLK) return hipErrorInvalidValue;
  const long long R = BH * T;
  {
    const int rows_per_block = 256 / WAVE;
    const long long grid = (R + rows_per_block - 1) / rows_per_block;
    hipLaunchKernelGGL(attn_delta_kernel, dim3(grid), dim3(256), 0, stream,
                       (const bf16*)dout, (const bf16*)o, delta, R);
  }
  dim3 grid(T / BLK, BH);
  hipLaunchKernelGGL(attn_bwd_dkv_kernel, grid, dim3(256), 0, stream,
                     (const bf16*)q, (const bf16*)k, (const bf16*)v,
                     (const bf16*)dout, lse, delta, (bf16*)dk, (bf16*)dv, T,
                     scale);
  hipLaunchKernelGGL(attn_bwd_dq_kernel, grid, dim3(256), 0, stream,
                     (const bf16*)q, (const bf16*)k, (const bf16*)v,
                     (const bf16*)dout, lse, delta, (bf16*)dq, T, scale);
  return hipGetLastError();
}

}  // extern "C"
