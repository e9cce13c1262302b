// Fused causal attention (flash-style) on CDNA4 MFMA matrix cores.
//
// Replaces the reference's materialized (B,H,T,T) score path
// (/root/reference/example/model.py:29-51) with an MI355X-native design
// built around v_mfma_f32_32x32x16_bf16 (fp32 accumulate):
//
//   forward : per Q block (NW waves x 32 rows), stream 64-key K/V tiles
//             through XOR-swizzled LDS. S^T = mfma(K, Q) puts the whole
//             softmax row in one lane pair (one shfl_xor(32) per reduce);
//             online softmax runs in exp2 space (log2e folded into the Q
//             prescale); P is repacked into MFMA A-fragments with TWO
//             v_permlane32_swap per k-slice (no LDS round trip); O = P V
//             with V consumed via ds_read_tr16_b64 hardware transpose
//             reads from the row-major image. Saves per-row natural-log lse.
//   backward: recompute-based two-kernel scheme (no atomics):
//             the dq kernel (launched first) owns a (NW*32)-row Q block,
//             computes delta = rowsum(dO*O) from its own fragments and
//             publishes it; the dkv kernel owns a (NW*32)-key block and
//             accumulates dK/dV over 64-row Q tiles, consuming delta.
//             The 1/sqrt(D) on dS is folded into the dK/dQ epilogue.
//
// All global accesses are stride-parameterized so the kernels consume the
// packed qkv projection layout (B,T,3,H,D) and write O/dQKV into (B,T,H,D)
// buffers directly — no transpose-copies on the hot path.
//
// VALU discipline (first profile measured 22:1 VALU:MFMA on a naive
// version): every LDS offset is precomputed before the K/V loop; causal
// masking runs only on diagonal tiles (wave-uniform branch); waves whose
// rows lie entirely outside a tile skip its compute.
//
// 32x32x16 fragment layouts (hardware-verified slot-pairing rule: A and B
// must agree on the slot->k map, C is fixed; see scripts/debug_mfma.py):
//   A[32x16]: lane l, elem e(0..7) -> A[l%32][(l/32)*8 + e]
//   B[16x32]: lane l, elem e      -> B[(l/32)*8 + e][l%32]
//   C[32x32]: lane l, reg r(0..15)-> C[(r&3) + 8*(r>>2) + 4*(l/32)][l%32]
//
// Contract: bf16, head_dim 64, T % 64 == 0 (the Python op falls back to a
// composite rocBLAS path otherwise, ops/attention.py).
#include "common.h"
#include "mfma.h"

#include <cstdlib>
#include <type_traits>

namespace tdsa {

constexpr int KVB = 64;   // keys / rows per LDS tile
constexpr int D = 64;     // head dim (all GPT-2 sizes)
constexpr float LOG2E = 1.4426950408889634f;
constexpr float LN2 = 0.6931471805599453f;

// Raw v_exp_f32: libm exp2f wraps every call in an ldexp+select sequence for
// |x| > 126 handling; our arguments are always <= 0 (score - running max /
// lse), where the bare instruction is exact-enough (1 ulp) and handles -inf
// -> 0. Measured 1:1 v_ldexp_f32 per exp in the softmax hot loop without it.
DEV_INLINE float fast_exp2(float x) { return __builtin_amdgcn_exp2f(x); }
// Raw v_rcp_f32 for the epilogue 1/l (IEEE div emits ~10-inst div_scale
// chains; the result feeds a bf16 store).
DEV_INLINE float fast_rcp(float x) { return __builtin_amdgcn_rcpf(x); }

// XCD-aware block remap (T1): the dispatcher places linear block b on XCD
// b % 8, and the x-dimension varies fastest, so the gridDim.x blocks that
// share one (batch, head) — and its K/V tiles — land on different XCDs'
// L2s. Remap so they share an XCD: within each window of 8*gridX linear
// ids, ids congruent mod 8 form one bh. Bijective when gridY % 8 == 0.
DEV_INLINE void xcd_remap(int gx, long long gy, int& tile, long long& bh) {
  if (gy % 8 == 0) {
    const long long id = bh * gx + tile;   // linear dispatch id
    tile = (int)((id % (8 * gx)) / 8);
    bh = (id / (8 * gx)) * 8 + (id % 8);
  }
}

// Per-tensor global strides in ELEMENTS (last dim contiguous).
struct GStride {
  long long b;
  long long h;
  int t;
};

// Counter-based dropout RNG: keyed by (bh, qrow, key) so the backward
// kernels REGENERATE the forward's mask from (seed, indices) — nothing is
// stored, matching the recompute design. 32-bit fold + murmur3 fmix32
// finalizer: the first splitmix64 version kept 64-bit temporaries live
// across the softmax loop and pushed the DROP template into a 312 B/lane
// spill — the avalanche quality of fmix32 is ample for dropout.
// keep <=> rng < threshold, threshold = (1-p) * 2^32.
DEV_INLINE bool drop_keep(unsigned long long seed, unsigned long long idx,
                          unsigned thr) {
  unsigned h = (unsigned)(idx ^ (idx >> 32));
  h += (unsigned)seed;
  h ^= (unsigned)(seed >> 32);
  h ^= h >> 16;
  h *= 0x85EBCA6Bu;
  h ^= h >> 13;
  h *= 0xC2B2AE35u;
  h ^= h >> 16;
  return h < thr;
}

// C-layout -> A-fragment repack. Values live per lane as 16 C registers per
// 32-wide tile (packed to bf16 word pairs wA[r1]=(r0=0,1), wB[r1]=(r0=2,3)).
// A-frag slice s (k = 16s + 8*h32 + e at this lane's own 32-axis index)
// takes word pairs from C registers r1 = 2(s&1)+h of BOTH halves: one
// permlane32_swap per word pair delivers own-half and partner-half at once.
struct PackedC {
  unsigned wA[2][4];  // [32-tile][r1]
  unsigned wB[2][4];
};

template <int S>
DEV_INLINE bfrag frag_from_packed(const PackedC& p) {
  const unsigned uA = p.wA[S >> 1][2 * (S & 1)];
  const unsigned vA = p.wA[S >> 1][2 * (S & 1) + 1];
  const unsigned uB = p.wB[S >> 1][2 * (S & 1)];
  const unsigned vB = p.wB[S >> 1][2 * (S & 1) + 1];
  auto rA = __builtin_amdgcn_permlane32_swap(uA, vA, false, false);
  auto rB = __builtin_amdgcn_permlane32_swap(uB, vB, false, false);
  union { bfrag f; unsigned w[4]; } r;
  r.w[0] = rA[0];
  r.w[1] = rB[0];
  r.w[2] = rA[1];
  r.w[3] = rB[1];
  return r.f;
}

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
template <int NW, bool DROP = false, int MINW = (NW == 8 ? 4 : 2)>
// NW=8: cap at 128 VGPR so two 8-wave blocks are resident per CU (at 132
// VGPR the 8-wave granularity rounds occupancy down to ONE block).
// DROP: fused attention dropout — the row sum l_run uses the UNMASKED
// exponentials (normalization is the true softmax sum) while the PV
// accumulation consumes masked/rescaled P.
__launch_bounds__(NW * WAVE, MINW)
__global__ void attn_fwd_kernel(const bf16* __restrict__ q,
                                const bf16* __restrict__ k,
                                const bf16* __restrict__ v,
                                bf16* __restrict__ o, float* __restrict__ lse,
                                int T, int H, float scale, GStride sq,
                                GStride so, unsigned long long seed,
                                unsigned keep_thr, float inv_keep) {
  constexpr int BM = NW * 32;
  constexpr int NT = NW * WAVE;
  __shared__ __attribute__((aligned(16))) char smem[2 * KVB * D * 2];
  char* lds_k = smem;                 // [64][64] keys row-major
  char* lds_v = smem + KVB * D * 2;   // [64][64] values row-major

  int qb = blockIdx.x;
  long long bh = blockIdx.y;
  xcd_remap(gridDim.x, gridDim.y, qb, bh);
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid / WAVE;
  const int q32 = lane & 31;
  const int h32 = lane >> 5;

  const long long boff = (bh / H) * sq.b + (bh % H) * sq.h;
  const bf16* qp = q + boff + (long long)(qb * BM) * sq.t;
  const bf16* kp = k + boff;
  const bf16* vp = v + boff;

  Stage<NT> stage_k(kp, tid, sq.t);
  Stage<NT> stage_v(vp, tid, sq.t);
  // K fragment offsets on the fly (same bit trick as AF_OFF in dkv)
  const int kf_row[2] = {q32 * 128, (32 + q32) * 128};
  const int kf_xk = (q32 & 7) << 4;
#define KF_OFF(t2, s) (kf_row[t2] + (((s) * 32 + 16 * h32) ^ kf_xk))
  int trb[2][2];
  tr_bases(lane, trb);
  const float qscale = scale * LOG2E;
  bfrag q_frag[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    q_frag[s] = load_frag_scaled(qp, w * 32 + q32, s * 16 + 8 * h32, sq.t,
                                 qscale);

  const f32x16 kzero = {};
  f32x16 o_acc[2] = {};
  float m_run = -INFINITY;
  float l_run = 0.f;

  const int row_lo = qb * BM + w * 32;
  const int row_me = row_lo + q32;
  // dropout index base (loop-invariant): idx = base + key
  const unsigned long long drop_base =
      DROP ? ((unsigned long long)bh * T + row_me) * T : 0;
  const int n_kv = (qb + 1) * BM / KVB;
  short8v pend_k[Stage<NT>::REPS], pend_v[Stage<NT>::REPS];
  stage_k.fetch(pend_k);
  stage_v.fetch(pend_v);
  for (int j = 0; j < n_kv; ++j) {
    __syncthreads();
    stage_k.put(lds_k, pend_k);
    stage_v.put(lds_v, pend_v);
    if (j + 1 < n_kv) {
      stage_k.advance();
      stage_v.advance();
      stage_k.fetch(pend_k);
      stage_v.fetch(pend_v);
    }
    __syncthreads();

    const int key0 = j * KVB;
    if (key0 > row_lo + 31) continue;

    // S^T tiles: C[key = 32*t2 + crow(r,h32)][qrow = q32], log2-scaled
    f32x16 st[2];
    __builtin_amdgcn_s_setprio(1);  // favor MFMA-issuing waves (T5)
#pragma unroll
    for (int t2 = 0; t2 < 2; ++t2) {
      // seed from the loop-invariant zero vector: a per-tile `= {}` init
      // re-emits 16 v_mov per accumulator per tile
      f32x16 acc = MFMA32(lds_read16(lds_k, KF_OFF(t2, 0)), q_frag[0], kzero);
#pragma unroll
      for (int s = 1; s < 4; ++s)
        acc = MFMA32(lds_read16(lds_k, KF_OFF(t2, s)), q_frag[s], acc);
      st[t2] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    const bool diag = key0 + KVB - 1 > row_lo;
    float mt = -INFINITY;
    if (diag) {
#pragma unroll
      for (int t2 = 0; t2 < 2; ++t2)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int key = key0 + t2 * 32 + crow(r, h32);
          float s = (key <= row_me) ? st[t2][r] : -INFINITY;
          st[t2][r] = s;
          mt = fmaxf(mt, s);
        }
    } else {
#pragma unroll
      for (int t2 = 0; t2 < 2; ++t2)
#pragma unroll
        for (int r = 0; r < 16; ++r) mt = fmaxf(mt, st[t2][r]);
    }
    mt = fmaxf(mt, __shfl_xor(mt, 32, WAVE));
    const float m_new = fmaxf(m_run, mt);
    const float alpha = fast_exp2(m_run - m_new);
    float psum = 0.f;
    PackedC P;
#pragma unroll
    for (int t2 = 0; t2 < 2; ++t2)
#pragma unroll
      for (int r1 = 0; r1 < 4; ++r1) {
        float p[4];
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          float pe = fast_exp2(st[t2][4 * r1 + e] - m_new);
          psum += pe;
          if (DROP) {
            const int key = key0 + t2 * 32 + crow(4 * r1 + e, h32);
            pe = drop_keep(seed, drop_base + key, keep_thr)
                     ? pe * inv_keep
                     : 0.f;
          }
          p[e] = pe;
        }
        P.wA[t2][r1] = pack2(p[0], p[1]);
        P.wB[t2][r1] = pack2(p[2], p[3]);
      }
    psum += __shfl_xor(psum, 32, WAVE);
    l_run = l_run * alpha + psum;
    m_run = m_new;

    if (__any(alpha != 1.0f)) {
      float a_row[16];
#pragma unroll
      for (int r = 0; r < 16; ++r)
        a_row[r] = __shfl(alpha, crow(r, h32), WAVE);
#pragma unroll
      for (int dt = 0; dt < 2; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[dt][r] *= a_row[r];
    }

    // PV
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      bfrag pa;
      switch (s) {
        case 0: pa = frag_from_packed<0>(P); break;
        case 1: pa = frag_from_packed<1>(P); break;
        case 2: pa = frag_from_packed<2>(P); break;
        default: pa = frag_from_packed<3>(P); break;
      }
#pragma unroll
      for (int dt = 0; dt < 2; ++dt)
        o_acc[dt] = MFMA32(pa, tr_bfrag(lds_v, trb[dt][0] + s * 2048,
                                        trb[dt][1] + s * 2048),
                           o_acc[dt]);
    }
    __builtin_amdgcn_s_setprio(0);
  }

  float linv_row[16];
#pragma unroll
  for (int r = 0; r < 16; ++r)
    linv_row[r] = fast_rcp(__shfl(l_run, crow(r, h32), WAVE));
  bf16* op = o + (bh / H) * so.b + (bh % H) * so.h
             + (long long)(qb * BM + w * 32) * so.t;
#pragma unroll
  for (int dt = 0; dt < 2; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r)
      op[crow(r, h32) * so.t + dt * 32 + q32] =
          f2bf(o_acc[dt][r] * linv_row[r]);
  if (lane < 32) {
    lse[bh * T + qb * BM + w * 32 + q32] = (m_run + log2f(l_run)) * LN2;
  }
}

// ---------------------------------------------------------------------------
// Backward dK/dV: one workgroup per (NW*32)-key block; wave w owns 32 keys.
// ---------------------------------------------------------------------------
template <int NW, bool DROP = false>
// unconstrained VGPRs: the compiler takes ~256 (2 waves/SIMD) — capping to
// 128 (MINW 4) spills 800 B/lane and 170 (MINW 3) still 288 B/lane, and the
// measured time at 256 equals the kernel's best (169.8 us at B8)
__launch_bounds__(NW * WAVE)
__global__ void attn_bwd_dkv_kernel(const bf16* __restrict__ q,
                                    const bf16* __restrict__ k,
                                    const bf16* __restrict__ v,
                                    const bf16* __restrict__ dout,
                                    const float* __restrict__ lse,
                                    const float* __restrict__ delta,
                                    bf16* __restrict__ dk, bf16* __restrict__ dv,
                                    int T, int H, float scale, GStride sq,
                                    GStride so, GStride sd,
                                    unsigned long long seed, unsigned keep_thr,
                                    float inv_keep) {
  constexpr int BK = NW * 32;
  constexpr int NT = NW * WAVE;
  __shared__ __attribute__((aligned(16))) char smem[2 * KVB * D * 2 + 2 * KVB * 4];
  char* lds_q = smem;
  char* lds_do = smem + KVB * D * 2;
  float* lds_lse = reinterpret_cast<float*>(smem + 2 * KVB * D * 2);
  float* lds_dlt = lds_lse + KVB;

  int jb = blockIdx.x;
  long long bh = blockIdx.y;
  xcd_remap(gridDim.x, gridDim.y, jb, bh);
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid / WAVE;
  const int k32 = lane & 31;   // this lane's key within the wave tile
  const int h32 = lane >> 5;

  const long long boff = (bh / H) * sq.b + (bh % H) * sq.h;
  const long long ooff = (bh / H) * so.b + (bh % H) * so.h;
  const bf16* qp = q + boff;
  const bf16* kp = k + boff;
  const bf16* vp = v + boff;
  const bf16* dop = dout + ooff;

  Stage<NT> stage_q(qp + (long long)(jb * BK / KVB * KVB) * sq.t, tid, sq.t);
  Stage<NT> stage_do(dop + (long long)(jb * BK / KVB * KVB) * so.t, tid, so.t);
  // fragment offsets on the fly: af(t2, s) = rowbase[t2] + (s-term ^ xk);
  // the s/h term and the swizzle XOR live in the same byte bits 4-6, so XOR
  // composes (keeps 8 VGPRs free for the staging split below)
  const int af_row[2] = {k32 * 128, (32 + k32) * 128};
  const int af_xk = (k32 & 7) << 4;
#define AF_OFF(t2, s) (af_row[t2] + (((s) * 32 + 16 * h32) ^ af_xk))
  int trb[2][2];
  tr_bases(lane, trb);
  const float kscale = scale * LOG2E;
  const int key_lo = jb * BK + w * 32;
  const int key_me = key_lo + k32;
  // dropout index base: idx = base + qrow*T + key_me (qrow*T fits 32 bits)
  const unsigned long long drop_bhTT =
      DROP ? (unsigned long long)bh * T * T : 0;
  bfrag k_frag[4], v_frag[4];  // B operands: col = k32, k-slices over D
#pragma unroll
  for (int s = 0; s < 4; ++s) {
    k_frag[s] = load_frag_scaled(kp, key_lo + k32, s * 16 + 8 * h32, sq.t,
                                 kscale);
    v_frag[s] = load_frag(vp, key_lo + k32, s * 16 + 8 * h32, sq.t);
  }

  const f32x16 kzero = {};
  f32x16 dk_acc[2] = {};
  f32x16 dv_acc[2] = {};

  const int i0 = jb * BK / KVB;
  short8v pend_q[Stage<NT>::REPS], pend_do[Stage<NT>::REPS];
  stage_q.fetch(pend_q);
  stage_do.fetch(pend_do);
  for (int i = i0; i < T / KVB; ++i) {
    __syncthreads();
    stage_q.put(lds_q, pend_q);
    stage_do.put(lds_do, pend_do);
    if (i + 1 < T / KVB) {
      stage_q.advance();
      stage_do.advance();
      stage_q.fetch(pend_q);   // next tile flies under this tile's compute
      stage_do.fetch(pend_do);
    }
    if (tid < KVB) {
      lds_lse[tid] = lse[bh * T + i * KVB + tid] * LOG2E;
      lds_dlt[tid] = delta[bh * T + i * KVB + tid];
    }
    __syncthreads();

    const int q0 = i * KVB;
    if (q0 + KVB - 1 < key_lo) continue;

    const bool diag = q0 < key_lo + 32;
    PackedC P, dS;
#pragma unroll
    for (int t2 = 0; t2 < 2; ++t2) {
      // S, dP tiles: C[qrow = 32*t2 + crow(r,h32)][key = k32]
      __builtin_amdgcn_s_setprio(1);
      f32x16 s_acc = MFMA32(lds_read16(lds_q, AF_OFF(t2, 0)), k_frag[0], kzero);
      f32x16 dp_acc = MFMA32(lds_read16(lds_do, AF_OFF(t2, 0)), v_frag[0], kzero);
#pragma unroll
      for (int s = 1; s < 4; ++s) {
        s_acc = MFMA32(lds_read16(lds_q, AF_OFF(t2, s)), k_frag[s], s_acc);
        dp_acc = MFMA32(lds_read16(lds_do, AF_OFF(t2, s)), v_frag[s], dp_acc);
      }
      __builtin_amdgcn_s_setprio(0);
      // MASK as a compile-time split: `diag` is wave-uniform but inside the
      // unrolled loop the single-version ternary was if-converted into
      // cmp+cndmask+add on EVERY tile (~96 VALU/tile measured in the .s);
      // only ~1 of 16 tiles is diagonal.
      auto consume = [&](auto MASK) {
#pragma unroll
        for (int r1 = 0; r1 < 4; ++r1) {
          float p[4], dsv[4];
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            const int r = 4 * r1 + e;
            const int lrow = t2 * 32 + crow(r, h32);
            float pp = fast_exp2(s_acc[r] - lds_lse[lrow]);
            if (MASK.value) pp = (key_me <= q0 + lrow) ? pp : 0.f;
            float pv = pp;                 // masked/rescaled P feeds dV
            float dpv = dp_acc[r];         // masked/rescaled dP feeds dS
            if (DROP) {
              const bool keep = drop_keep(
                  seed,
                  drop_bhTT + (unsigned)((q0 + lrow) * T) + key_me,
                  keep_thr);
              pv = keep ? pp * inv_keep : 0.f;
              dpv = keep ? dpv * inv_keep : 0.f;
            }
            p[e] = pv;
            dsv[e] = pp * (dpv - lds_dlt[lrow]);  // scale in epilogue
          }
          P.wA[t2][r1] = pack2(p[0], p[1]);
          P.wB[t2][r1] = pack2(p[2], p[3]);
          dS.wA[t2][r1] = pack2(dsv[0], dsv[1]);
          dS.wB[t2][r1] = pack2(dsv[2], dsv[3]);
        }
      };
      if (diag)
        consume(std::true_type{});
      else
        consume(std::false_type{});
    }

    // dV[key][d] += P^T dO ; dK[key][d] += dS^T Q  (A-frag k = qrow)
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      bfrag ap, as;
      switch (s) {
        case 0: ap = frag_from_packed<0>(P); as = frag_from_packed<0>(dS); break;
        case 1: ap = frag_from_packed<1>(P); as = frag_from_packed<1>(dS); break;
        case 2: ap = frag_from_packed<2>(P); as = frag_from_packed<2>(dS); break;
        default: ap = frag_from_packed<3>(P); as = frag_from_packed<3>(dS); break;
      }
#pragma unroll
      for (int dt = 0; dt < 2; ++dt) {
        dv_acc[dt] = MFMA32(ap, tr_bfrag(lds_do, trb[dt][0] + s * 2048,
                                         trb[dt][1] + s * 2048),
                            dv_acc[dt]);
        dk_acc[dt] = MFMA32(as, tr_bfrag(lds_q, trb[dt][0] + s * 2048,
                                         trb[dt][1] + s * 2048),
                            dk_acc[dt]);
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

  const long long doff = (bh / H) * sd.b + (bh % H) * sd.h
                         + (long long)(key_lo) * sd.t;
  bf16* dkp = dk + doff;
  bf16* dvp = dv + doff;
#pragma unroll
  for (int dt = 0; dt < 2; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      dkp[crow(r, h32) * sd.t + dt * 32 + k32] = f2bf(dk_acc[dt][r] * scale);
      dvp[crow(r, h32) * sd.t + dt * 32 + k32] = f2bf(dv_acc[dt][r]);
    }
}

// ---------------------------------------------------------------------------
// Backward dQ: one workgroup per (NW*32)-row Q block; wave w owns 32 rows.
// ---------------------------------------------------------------------------
template <int NW, bool DROP = false>
__launch_bounds__(NW * WAVE)  // capping at 128 VGPR spills 164 B/lane here
__global__ void attn_bwd_dq_kernel(const bf16* __restrict__ q,
                                   const bf16* __restrict__ k,
                                   const bf16* __restrict__ v,
                                   const bf16* __restrict__ o,
                                   const bf16* __restrict__ dout,
                                   const float* __restrict__ lse,
                                   float* __restrict__ delta,
                                   bf16* __restrict__ dq, int T, int H,
                                   float scale, GStride sq, GStride so,
                                   GStride sd, unsigned long long seed,
                                   unsigned keep_thr, float inv_keep) {
  constexpr int BM = NW * 32;
  constexpr int NT = NW * WAVE;
  __shared__ __attribute__((aligned(16))) char smem[2 * KVB * D * 2];
  char* lds_k = smem;                     // K row-major
  char* lds_v = smem + KVB * D * 2;       // V row-major

  int qb = blockIdx.x;
  long long bh = blockIdx.y;
  xcd_remap(gridDim.x, gridDim.y, qb, bh);
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid / WAVE;
  const int q32 = lane & 31;
  const int h32 = lane >> 5;

  const long long boff = (bh / H) * sq.b + (bh % H) * sq.h;
  const long long ooff = (bh / H) * so.b + (bh % H) * so.h;
  const bf16* qp = q + boff + (long long)(qb * BM) * sq.t;
  const bf16* kp = k + boff;
  const bf16* vp = v + boff;
  const bf16* dop = dout + ooff + (long long)(qb * BM) * so.t;

  Stage<NT> stage_k(kp, tid, sq.t);
  Stage<NT> stage_v(vp, tid, sq.t);
  int f_off[2][4];
#pragma unroll
  for (int t2 = 0; t2 < 2; ++t2)
#pragma unroll
    for (int s = 0; s < 4; ++s)
      f_off[t2][s] = swz(t2 * 32 + q32, (s * 16 + 8 * h32) * 2);
  int trb[2][2];
  tr_bases(lane, trb);
  const float qscale = scale * LOG2E;
  bfrag q_frag[4], do_frag[4];
#pragma unroll
  for (int s = 0; s < 4; ++s) {
    q_frag[s] = load_frag_scaled(qp, w * 32 + q32, s * 16 + 8 * h32, sq.t,
                                 qscale);
    do_frag[s] = load_frag(dop, w * 32 + q32, s * 16 + 8 * h32, so.t);
  }
  const int row_lo = qb * BM + w * 32;
  const int row_me = row_lo + q32;
  const unsigned long long drop_base =
      DROP ? ((unsigned long long)bh * T + row_me) * T : 0;
  const float lse2_me = lse[bh * T + row_me] * LOG2E;
  // delta = rowsum(dO*O) computed here from the fragments already in
  // registers (+ one O load) and PUBLISHED for the dkv kernel, which is
  // launched after this one — replaces a separate full-pass delta kernel.
  float dlt_me = 0.f;
#pragma unroll
  for (int s = 0; s < 4; ++s) {
    union { bfrag f; short8v v8; } ov;
    ov.f = load_frag(o + ooff + (long long)(qb * BM) * so.t, w * 32 + q32,
                     s * 16 + 8 * h32, so.t);
    union { bfrag f; short8v v8; } dv;
    dv.f = do_frag[s];
#pragma unroll
    for (int e = 0; e < 8; ++e)
      dlt_me += bf_elem(ov.v8, e) * bf_elem(dv.v8, e);
  }
  dlt_me += __shfl_xor(dlt_me, 32, WAVE);
  if (lane < 32) delta[bh * T + row_me] = dlt_me;

  const f32x16 kzero = {};
  f32x16 dq_acc[2] = {};

  const int n_kv = (qb + 1) * BM / KVB;
  short8v pend_k[Stage<NT>::REPS], pend_v[Stage<NT>::REPS];
  stage_k.fetch(pend_k);
  stage_v.fetch(pend_v);
  for (int j = 0; j < n_kv; ++j) {
    __syncthreads();
    stage_k.put(lds_k, pend_k);
    stage_v.put(lds_v, pend_v);
    if (j + 1 < n_kv) {
      stage_k.advance();
      stage_v.advance();
      stage_k.fetch(pend_k);   // next tile flies under this tile's compute
      stage_v.fetch(pend_v);
    }
    __syncthreads();

    const int key0 = j * KVB;
    if (key0 > row_lo + 31) continue;

    const bool diag = key0 + KVB - 1 > row_lo;
    PackedC dS;
#pragma unroll
    for (int t2 = 0; t2 < 2; ++t2) {
      // S^T, dP^T tiles: C[key = 32*t2 + crow(r,h32)][qrow = q32]
      __builtin_amdgcn_s_setprio(1);
      f32x16 s_acc = MFMA32(lds_read16(lds_k, f_off[t2][0]), q_frag[0], kzero);
      f32x16 dp_acc = MFMA32(lds_read16(lds_v, f_off[t2][0]), do_frag[0], kzero);
#pragma unroll
      for (int s = 1; s < 4; ++s) {
        s_acc = MFMA32(lds_read16(lds_k, f_off[t2][s]), q_frag[s], s_acc);
        dp_acc = MFMA32(lds_read16(lds_v, f_off[t2][s]), do_frag[s], dp_acc);
      }
      __builtin_amdgcn_s_setprio(0);
      float dsv[16];
      // wave-uniform mask split (same rationale as the dkv kernel)
      auto consume = [&](auto MASK) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float pp = fast_exp2(s_acc[r] - lse2_me);
          if (MASK.value) {
            const int key = key0 + t2 * 32 + crow(r, h32);
            pp = (key <= row_me) ? pp : 0.f;
          }
          float dpv = dp_acc[r];
          if (DROP) {
            const int key = key0 + t2 * 32 + crow(r, h32);
            dpv = drop_keep(seed, drop_base + key, keep_thr)
                      ? dpv * inv_keep
                      : 0.f;
          }
          dsv[r] = pp * (dpv - dlt_me);
        }
      };
      if (diag)
        consume(std::true_type{});
      else
        consume(std::false_type{});
#pragma unroll
      for (int r1 = 0; r1 < 4; ++r1) {
        dS.wA[t2][r1] = pack2(dsv[4 * r1], dsv[4 * r1 + 1]);
        dS.wB[t2][r1] = pack2(dsv[4 * r1 + 2], dsv[4 * r1 + 3]);
      }
    }

    // dQ[qrow][d] += dS K  (A-frag k = key)
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      bfrag as;
      switch (s) {
        case 0: as = frag_from_packed<0>(dS); break;
        case 1: as = frag_from_packed<1>(dS); break;
        case 2: as = frag_from_packed<2>(dS); break;
        default: as = frag_from_packed<3>(dS); break;
      }
#pragma unroll
      for (int dt = 0; dt < 2; ++dt)
        dq_acc[dt] = MFMA32(as, tr_bfrag(lds_k, trb[dt][0] + s * 2048,
                                         trb[dt][1] + s * 2048),
                            dq_acc[dt]);
    }
    __builtin_amdgcn_s_setprio(0);
  }

  bf16* dqp = dq + (bh / H) * sd.b + (bh % H) * sd.h
              + (long long)(qb * BM + w * 32) * sd.t;
#pragma unroll
  for (int dt = 0; dt < 2; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r)
      dqp[crow(r, h32) * sd.t + dt * 32 + q32] = f2bf(dq_acc[dt][r] * scale);
}

}  // namespace tdsa

using namespace tdsa;

extern "C" {

// Within-box tuning knobs (defaults are the measured-best dispatch; the env
// override exists for A/B because box-to-box wall-clock noise is ~3%).
static int env_nw(const char* name, int dflt) {
  const char* v = getenv(name);
  if (!v) return dflt;
  int n = atoi(v);
  return (n == 2 || n == 4 || n == 8) ? n : dflt;
}

// strides arrays: {b, h, t} in elements, per tensor group:
// sq = q/k/v, so = o (and dO in bwd), sd = dq/dk/dv.
// dropout_p > 0 enables the fused-dropout template (seed regenerates the
// identical mask in the backward kernels).
hipError_t tdsa_attn_fwd(const void* q, const void* k, const void* v, void* o,
                         float* lse, long long B, long long H, int T,
                         float scale, const long long* sq_in,
                         const long long* so_in, float dropout_p,
                         unsigned long long seed, hipStream_t stream) {
  if (T % KVB) return hipErrorInvalidValue;
  GStride sq{sq_in[0], sq_in[1], (int)sq_in[2]};
  GStride so{so_in[0], so_in[1], (int)so_in[2]};
  const bool drop = dropout_p > 0.0f;
  const double keepd = 1.0 - (double)dropout_p;
  // clamp: for p -> 0, keepd*2^32 == 2^32 overflows (unsigned) to 0 and
  // would drop EVERYTHING instead of nothing
  const double thr_d = keepd * 4294967296.0;
  const unsigned keep_thr =
      thr_d >= 4294967295.0 ? 0xFFFFFFFFu : (unsigned)thr_d;
  const float inv_keep = (float)(1.0 / keepd);
#define LAUNCH_FWD(NW, MINW)                                                  \
  do {                                                                        \
    if (drop)                                                                 \
      hipLaunchKernelGGL((attn_fwd_kernel<NW, true, MINW>),                   \
                         dim3(T / (NW * 32), B * H), dim3(NW * WAVE), 0,      \
                         stream, (const bf16*)q, (const bf16*)k,              \
                         (const bf16*)v, (bf16*)o, lse, T, (int)H, scale,     \
                         sq, so, seed, keep_thr, inv_keep);                   \
    else                                                                      \
      hipLaunchKernelGGL((attn_fwd_kernel<NW, false, MINW>),                  \
                         dim3(T / (NW * 32), B * H), dim3(NW * WAVE), 0,      \
                         stream, (const bf16*)q, (const bf16*)k,              \
                         (const bf16*)v, (bf16*)o, lse, T, (int)H, scale,     \
                         sq, so, seed, keep_thr, inv_keep);                   \
  } while (0)
  const int nw_fwd = env_nw("TDSA_ATTN_FWD_NW", 8);
  const char* mv = getenv("TDSA_ATTN_FWD_MINW");
  const bool relaxed = (mv && atoi(mv) == 3) || drop;
  // dropout: the no-drop kernel already fills the 128-VGPR cap, so the
  // DROP variant spills 316 B/lane under MINW=4 — the relaxed cap
  // (168 VGPR, 1 block/CU) measured faster for it
  if (T % 256 == 0 && nw_fwd == 8) {
    if (relaxed) LAUNCH_FWD(8, 3);  // A/B: no 128-cap spill, 1 block/CU
    else LAUNCH_FWD(8, 4);
  }
  else if (T % 128 == 0 && nw_fwd >= 4) LAUNCH_FWD(4, 2);
  else LAUNCH_FWD(2, 2);
#undef LAUNCH_FWD
  return hipGetLastError();
}

hipError_t tdsa_attn_bwd(const void* q, const void* k, const void* v,
                         const void* o, const float* lse, const void* dout,
                         void* dq, void* dk, void* dv, float* delta,
                         long long B, long long H, int T, float scale,
                         const long long* sq_in, const long long* so_in,
                         const long long* sd_in, float dropout_p,
                         unsigned long long seed, hipStream_t stream) {
  if (T % KVB) return hipErrorInvalidValue;
  GStride sq{sq_in[0], sq_in[1], (int)sq_in[2]};
  GStride so{so_in[0], so_in[1], (int)so_in[2]};
  GStride sd{sd_in[0], sd_in[1], (int)sd_in[2]};
  const long long BH = B * H;
  const bool drop = dropout_p > 0.0f;
  const double keepd = 1.0 - (double)dropout_p;
  // clamp: for p -> 0, keepd*2^32 == 2^32 overflows (unsigned) to 0 and
  // would drop EVERYTHING instead of nothing
  const double thr_d = keepd * 4294967296.0;
  const unsigned keep_thr =
      thr_d >= 4294967295.0 ? 0xFFFFFFFFu : (unsigned)thr_d;
  const float inv_keep = (float)(1.0 / keepd);
  // dq runs FIRST: it computes and publishes delta = rowsum(dO*O) from
  // fragments it loads anyway; dkv (same stream) consumes it.
#define LAUNCH_BWD_D(NW, DR)                                                  \
  do {                                                                        \
    dim3 grid(T / (NW * 32), BH);                                             \
    hipLaunchKernelGGL((attn_bwd_dq_kernel<NW, DR>), grid, dim3(NW * WAVE),   \
                       0, stream, (const bf16*)q, (const bf16*)k,             \
                       (const bf16*)v, (const bf16*)o, (const bf16*)dout,     \
                       lse, delta, (bf16*)dq, T, (int)H, scale, sq, so, sd,   \
                       seed, keep_thr, inv_keep);                             \
    hipLaunchKernelGGL((attn_bwd_dkv_kernel<NW, DR>), grid, dim3(NW * WAVE),  \
                       0, stream, (const bf16*)q, (const bf16*)k,             \
                       (const bf16*)v, (const bf16*)dout, lse, delta,         \
                       (bf16*)dk, (bf16*)dv, T, (int)H, scale, sq, so, sd,    \
                       seed, keep_thr, inv_keep);                             \
  } while (0)
#define LAUNCH_BWD(NW)                                                        \
  do {                                                                        \
    if (drop) LAUNCH_BWD_D(NW, true);                                         \
    else LAUNCH_BWD_D(NW, false);                                             \
  } while (0)
  // 8-wave blocks measured 285us vs 332us for 4-wave at B8/H16/T1024
  // (pre-tr16); staging amortization across 8 waves beat block overlap.
  const int nw_bwd = env_nw("TDSA_ATTN_BWD_NW", 8);
  if (T % 256 == 0 && nw_bwd == 8) LAUNCH_BWD(8);
  else if (T % 128 == 0 && nw_bwd >= 4) LAUNCH_BWD(4);
  else LAUNCH_BWD(2);
#undef LAUNCH_BWD
  return hipGetLastError();
}

}  // extern "C"
