// Hand-written CDNA4 TN GEMM for the Linear weight gradient:
//
//   dw[N,K] = dy[M,N]^T @ x[M,K]     (bf16 in, fp32 accumulate)
//
// The ONE GEMM shape family where the reduction dim (M = B*T, 32k typical)
// is the ROW dim of BOTH row-major operands. That makes both MFMA operands
// reachable through the same ds_read_tr16_b64 hardware-transpose read of
// row-major staged [64][64] LDS images (mfma.h: the A and B fragment
// layouts of v_mfma_f32_32x32x16_bf16 share one lane->(k-elem, 32-axis)
// map) — no transposed staging, no bank conflicts, no strided global reads.
// hipBLASLt's weakest library shape per profiles/gemm_shapes_hipblaslt.txt;
// registered as the autotuner's first candidate for linear_dw
// (ops/linear.py), so the library wins any shape where it is faster.
//
// Geometry: 4 waves / workgroup, 128(N) x 128(K) output tile; wave w owns
// the 32-row n-band [32w, 32w+32) x 128 k. Per 64-row m-chunk each wave
// issues 4 A tr-reads + 16 B tr-reads + 16 MFMAs from four 16 KB swizzled
// LDS images (dy 64x128, x 64x128), with the next chunk's global loads
// issued before the compute (fetch/put split — they fly under the MFMAs).
// The output is tiny (N*K vs M*K inputs), so workgroups additionally
// split the M reduction (grid.y) and combine 128x128 fp32 partials with
// atomicAdd into a zeroed fp32 buffer (skipped when grid.y == 1); the
// caller converts to bf16.
//
// Reference hot path being replaced:
// /root/reference/tiny_deepspeed/core/module/ops/linear.py:59-68.
#include "common.h"
#include "mfma.h"

#include <cstdlib>

namespace tdsa {

// XCD-group remap: the dispatcher places linear block b on XCD b % 8 with
// x varying fastest, so the tiles_k workgroups that share one (tn, split)
// — and therefore stream the SAME dy column-slice — would land on 8
// different XCDs' L2s and fetch it 8x from HBM. Remap so they share an
// XCD. Bijective when tiles_n * splits % 8 == 0; identity otherwise.
DEV_INLINE void tn_remap(int gx, int gy, int& tile, int& split, int tiles_k) {
  const int groups = (gx / tiles_k) * gy;  // (tn, split) pairs
  if (groups % 8) return;
  const long long id = (long long)split * gx + tile;
  const long long window = 8LL * tiles_k;
  const int g = (int)((id / window) * 8 + (id % 8));
  const int tk = (int)((id % window) / 8);
  const int tn = g / gy;
  split = g % gy;
  tile = tn * tiles_k + tk;
}

// Supertile remap (mode 2): group tiles into 2(tn) x 4(tk) rectangles and
// place each rectangle's 8 tiles on ONE XCD (ids congruent mod 8 within a
// 64-id window) — the XCD's L2 then re-serves both dy slices (2) and x
// slices (4) instead of streaming 8 distinct slice pairs. Needs
// tiles_n % 2 == 0, tiles_k % 4 == 0 and (supertiles * splits) % 8 == 0.
DEV_INLINE void tn_remap2(int gx, int gy, int& tile, int& split,
                          int tiles_k) {
  const int tiles_n = gx / tiles_k;
  if (tiles_n % 2 || tiles_k % 4) return;
  const int sn = tiles_n >> 1, sk = tiles_k >> 2;
  if ((sn * sk * gy) % 8) return;
  const long long id = (long long)split * gx + tile;
  const long long w = id >> 6;        // 64-id window
  const int s_local = (int)(id & 7);  // supertile within window (XCD)
  const int t_local = (int)((id & 63) >> 3);  // tile within supertile
  const long long sup = w * 8 + s_local;
  const int per_split = sn * sk;
  split = (int)(sup / per_split);
  const int ss = (int)(sup % per_split);
  const int tn = (ss / sk) * 2 + (t_local >> 2);
  const int tk = (ss % sk) * 4 + (t_local & 3);
  tile = tn * tiles_k + tk;
}

// NW waves; output tile (NW*32) x 128. Wave w owns n-band [32w, 32w+32) x
// 128 k. NW=8 halves the staged bytes per FLOP vs NW=4 (the dy slice is
// amortized over twice the MFMA work) — used when N % 256 == 0.
template <int NW, bool ATOMIC>
__launch_bounds__(NW * WAVE)
__global__ void gemm_tn_kernel(const bf16* __restrict__ dy,
                               const bf16* __restrict__ x,
                               float* __restrict__ dw, long long M, int N,
                               int K, long long m_per_split, int remap) {
  const int tiles_k = K >> 7;
  int tile = blockIdx.x;
  int split = blockIdx.y;
  if (remap == 2)
    tn_remap2(gridDim.x, gridDim.y, tile, split, tiles_k);
  else if (remap)
    tn_remap(gridDim.x, gridDim.y, tile, split, tiles_k);
  const int tn = tile / tiles_k;
  const int tk = tile % tiles_k;
  const int n0 = tn * (NW * 32);
  const int k0 = tk << 7;
  const long long m_begin = (long long)split * m_per_split;
  long long m_end = m_begin + m_per_split;
  if (m_end > M) m_end = M;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wave = tid >> 6;
  const int h32 = lane >> 5;

  constexpr int NT = NW * WAVE;
  constexpr int NIMG = NW / 2 + 2;         // dy images + 2 x images
  constexpr int IMG = 64 * 128;            // one swizzled [64][64] tile, 8 KB
  __shared__ char lds[NIMG * IMG];

  // images 0..NW/2-1: dy 64-col slices; images NW/2, NW/2+1: x lo/hi
  Stage<NT> st[NIMG];
#pragma unroll
  for (int i = 0; i < NW / 2; ++i)
    st[i].init(dy + m_begin * N + n0 + 64 * i, tid, N);
  st[NW / 2].init(x + m_begin * K + k0, tid, K);
  st[NW / 2 + 1].init(x + m_begin * K + k0 + 64, tid, K);

  int trb[2][2];
  tr_bases(lane, trb);
  const char* a_img = lds + (wave >> 1) * IMG;
  const int dta = wave & 1;

  f32x16 acc[4];
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc[dt][r] = 0.f;

  const int nchunks = m_end > m_begin ? (int)((m_end - m_begin) >> 6) : 0;
  if (nchunks > 0) {
#pragma unroll
    for (int i = 0; i < NIMG; ++i) {
      st[i].run(lds + i * IMG);
      st[i].advance();
    }
  }
  for (int c = 0; c < nchunks; ++c) {
    __syncthreads();
    short8v pend[NIMG][Stage<NT>::REPS];
    const bool more = c + 1 < nchunks;
    if (more) {
#pragma unroll
      for (int i = 0; i < NIMG; ++i) st[i].fetch(pend[i]);
    }
#pragma unroll
    for (int S = 0; S < 4; ++S) {
      const bfrag a = tr_bfrag(a_img, trb[dta][0] + S * 2048,
                               trb[dta][1] + S * 2048);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        const bfrag b = tr_bfrag(lds + (NW / 2 + (dt >> 1)) * IMG,
                                 trb[dt & 1][0] + S * 2048,
                                 trb[dt & 1][1] + S * 2048);
        acc[dt] = MFMA32(a, b, acc[dt]);
      }
    }
    __syncthreads();
    if (more) {
#pragma unroll
      for (int i = 0; i < NIMG; ++i) {
        st[i].put(lds + i * IMG, pend[i]);
        st[i].advance();
      }
    }
  }

  // epilogue: C[32x32] lane l reg r -> row crow(r,h32), col l%32
  const int col = k0 + (lane & 31);
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const long long idx =
          (long long)(n0 + wave * 32 + crow(r, h32)) * K + col + dt * 32;
      if (ATOMIC)
        atomicAdd(dw + idx, acc[dt][r]);
      else
        dw[idx] = acc[dt][r];
    }
  }
}

}  // namespace tdsa

using namespace tdsa;

extern "C" {

// Tile height: 4 waves / 128 rows measured BEST (the 8-wave 256-row tile
// variant lost 10-20% on every shape despite doubled staging intensity —
// gpurun gemm_tn_v2 vs v1 sweeps); 8 kept behind the env for re-testing.
static int tn_nw(int N) {
  int nw = 4;
  if (const char* v = getenv("TDSA_GEMM_TN_NW")) {
    int e = atoi(v);
    if ((e == 4 || e == 8) && N % (e * 32) == 0) nw = e;
  }
  return nw;
}

// Returns the split count the launcher will use (for the caller to decide
// whether the fp32 buffer needs zeroing); <=0 means shape unsupported.
int tdsa_gemm_tn_splits(long long M, int N, int K) {
  if (M <= 0 || (M & 63) || (N & 127) || (K & 127)) return 0;
  const int nw = tn_nw(N);
  const long long tiles = (long long)(N / (nw * 32)) * (K >> 7);
  // ~512 total workgroups (2 per CU): the splits sweep measured 8 splits
  // at 64 tiles (512 wgs) 25% faster than 32 splits — more splits just
  // multiply the fp32 atomic traffic once every CU has work
  long long want = (512 + tiles - 1) / tiles;
  long long maxs = M >> 6;  // at least one 64-chunk per split
  if (want > maxs) want = maxs;
  if (const char* v = getenv("TDSA_GEMM_TN_SPLITS")) want = atoll(v);
  if (want < 1) want = 1;
  return (int)want;
}

hipError_t tdsa_gemm_tn(const void* dy, const void* x, float* dw, long long M,
                        int N, int K, hipStream_t stream) {
  const int splits = tdsa_gemm_tn_splits(M, N, K);
  if (splits <= 0) return hipErrorInvalidValue;
  const int nw = tn_nw(N);
  const long long chunks_per_split = ((M >> 6) + splits - 1) / splits;
  const long long m_per_split = chunks_per_split << 6;
  // mode 2 (supertile) measured +1-5% across shapes; mode 1 (tn-group)
  // neutral — the chip-wide Infinity Cache already serves shared slices,
  // so XCD placement only trims the L2-miss tail
  int remap = 2;
  if (const char* v = getenv("TDSA_GEMM_TN_REMAP")) remap = atoi(v);
  dim3 grid((N / (nw * 32)) * (K >> 7), splits);
#define TN_LAUNCH(NWV, AT)                                                  \
  hipLaunchKernelGGL((gemm_tn_kernel<NWV, AT>), grid, dim3(NWV * WAVE), 0,  \
                     stream, (const bf16*)dy, (const bf16*)x, dw, M, N, K,  \
                     m_per_split, remap)
  if (nw == 8) {
    if (splits == 1) TN_LAUNCH(8, false); else TN_LAUNCH(8, true);
  } else {
    if (splits == 1) TN_LAUNCH(4, false); else TN_LAUNCH(4, true);
  }
#undef TN_LAUNCH
  return hipGetLastError();
}

}  // extern "C"
