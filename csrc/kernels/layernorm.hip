// LayerNorm forward / backward CDNA4 kernels (last-dim, affine+bias).
//
// Replaces the reference's three Triton kernels
// (/root/reference/tiny_deepspeed/core/module/ops/layernorm.py:158-298) with
// an MI355X-native design: one workgroup per row (grid-strided), fp32
// accumulation, bf16x8 / float4 vectorized global traffic, and a
// conflict-free two-pass dw/db reduction through per-block fp32 stripe
// buffers instead of the reference's spin-lock atomic_cas scheme
// (SURVEY.md 2.10A).
#include "common.h"

#include <cstdlib>

namespace tdsa {

template <typename T> struct VecTraits;
template <> struct VecTraits<float> {
  static constexpr int W = 4;
  typedef float4v V;
  static DEV_INLINE V load(const float* p) { return *reinterpret_cast<const V*>(p); }
  static DEV_INLINE float get(V v, int i) { return v[i]; }
  static DEV_INLINE void set(V& v, int i, float x) { v[i] = x; }
  static DEV_INLINE void store(float* p, V v) { *reinterpret_cast<V*>(p) = v; }
};
template <> struct VecTraits<bf16> {
  static constexpr int W = 8;
  typedef short8v V;
  static DEV_INLINE V load(const bf16* p) { return load8(p); }
  static DEV_INLINE float get(V v, int i) { return bf_elem(v, i); }
  static DEV_INLINE void set(V& v, int i, float x) { v[i] = bf_pack(x); }
  static DEV_INLINE void store(bf16* p, V v) { store8(p, v); }
};

// HAS_RES: fused residual add — h = x + res is computed in-kernel, written
// out (the residual stream) and normalized, saving the separate elementwise
// add pass + launch per LayerNorm site (2 per transformer block).
// ONE_WAVE: 64-thread blocks, one wave per row — the two per-row reductions
// become pure wave shuffles with NO __syncthreads / LDS at all (needs a
// deeper register cache: MAXITER up to 4 covers N<=2048 bf16 / 1024 fp32).
template <typename T, bool HAS_RES, int MAXITER = 2, bool ONE_WAVE = false>
__global__ void ln_fwd_kernel(const T* __restrict__ x, const T* __restrict__ res,
                              T* __restrict__ h, const T* __restrict__ w,
                              const T* __restrict__ b, T* __restrict__ y,
                              float* __restrict__ mean, float* __restrict__ rstd,
                              int M, int N, float eps) {
  using VT = VecTraits<T>;
  constexpr int W = VT::W;
  __shared__ float scratch[2 * 1024 / WAVE];
  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  for (int row = blockIdx.x; row < M; row += gridDim.x) {
    const T* xr = x + (long long)row * N;
    const T* rr = HAS_RES ? res + (long long)row * N : nullptr;
    T* hr = HAS_RES ? h + (long long)row * N : nullptr;
    typename VT::V xc[MAXITER];
    float s = 0.f, ss = 0.f;
    {
      int it = 0;
      for (int i = tid * W; i < N; i += nth * W, ++it) {
        typename VT::V v = VT::load(xr + i);
        if (HAS_RES) {
          typename VT::V rv = VT::load(rr + i);
          typename VT::V hv;
#pragma unroll
          for (int k = 0; k < W; ++k)
            VT::set(hv, k, VT::get(v, k) + VT::get(rv, k));
          VT::store(hr + i, hv);
          v = hv;
        }
        xc[it] = v;
#pragma unroll
        for (int k = 0; k < W; ++k) {
          float f = VT::get(v, k);
          s += f;
          ss += f * f;
        }
      }
    }
    if (ONE_WAVE) {
      s = wave_sum(s);
      ss = wave_sum(ss);
    } else {
      block_sum2(s, ss, scratch);
    }
    const float mu = s / N;
    const float var = fmaxf(ss / N - mu * mu, 0.0f);
    const float rs = rsqrtf(var + eps);
    if (tid == 0) {
      mean[row] = mu;
      rstd[row] = rs;
    }
    T* yr = y + (long long)row * N;
    {
      int it = 0;
      for (int i = tid * W; i < N; i += nth * W, ++it) {
        typename VT::V xv = xc[it];
        typename VT::V wv = VT::load(w + i);
        typename VT::V bv = VT::load(b + i);
        typename VT::V ov;
#pragma unroll
        for (int k = 0; k < W; ++k) {
          float xh = (VT::get(xv, k) - mu) * rs;
          VT::set(ov, k, xh * VT::get(wv, k) + VT::get(bv, k));
        }
        VT::store(yr + i, ov);
      }
    }
  }
}

// Backward dx (+ optionally the dw/db stripe partials). Grid = G blocks;
// block g handles rows g, g+G, ...; with STRIPES it accumulates dw/db into
// pdw[g*N..] (no atomics).
// Measured split (r2): the fused form chains rows with two block-wide
// reduction barriers each, capping dx at ~2.1 TB/s while the fwd kernel
// reaches 3.1. The launcher now runs dx at FULL grid (one row per block,
// STRIPES=false) and a separate barrier-free streaming kernel
// (ln_dwdb_accum_kernel) produces the stripes — one extra dy/x read buys
// unchained dx rows.
// HAS_DH: dx += dh (gradient of the residual-stream output h), fusing the
// backward-side elementwise add of the residual connection.
template <typename T, int MAXITER, bool HAS_DH, bool STRIPES = true,
          bool ONE_WAVE = false>
__global__ void ln_bwd_dx_kernel(const T* __restrict__ dy, const T* __restrict__ dh,
                                 const T* __restrict__ x,
                                 const T* __restrict__ w, const float* __restrict__ mean,
                                 const float* __restrict__ rstd, T* __restrict__ dx,
                                 float* __restrict__ pdw, float* __restrict__ pdb,
                                 int M, int N) {
  using VT = VecTraits<T>;
  constexpr int W = VT::W;
  __shared__ float scratch[2 * 1024 / WAVE];
  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  float accdw[MAXITER][W];
  float accdb[MAXITER][W];
#pragma unroll
  for (int it = 0; it < MAXITER; ++it)
#pragma unroll
    for (int k = 0; k < W; ++k) accdw[it][k] = accdb[it][k] = 0.f;

  for (int row = blockIdx.x; row < M; row += gridDim.x) {
    const T* dyr = dy + (long long)row * N;
    const T* xr = x + (long long)row * N;
    const float mu = mean[row];
    const float rs = rstd[row];
    typename VT::V dyc[MAXITER], xc[MAXITER];
    float c1 = 0.f, c2 = 0.f;
    {
      int it = 0;
      for (int i = tid * W; i < N; i += nth * W, ++it) {
        typename VT::V dv = VT::load(dyr + i);
        typename VT::V xv = VT::load(xr + i);
        typename VT::V wv = VT::load(w + i);
        dyc[it] = dv;
        xc[it] = xv;
#pragma unroll
        for (int k = 0; k < W; ++k) {
          float d = VT::get(dv, k);
          float xh = (VT::get(xv, k) - mu) * rs;
          float wdy = VT::get(wv, k) * d;
          c1 += xh * wdy;
          c2 += wdy;
          if (STRIPES) {
            accdw[it][k] += d * xh;
            accdb[it][k] += d;
          }
        }
      }
    }
    if (ONE_WAVE) {
      c1 = wave_sum(c1);
      c2 = wave_sum(c2);
    } else {
      block_sum2(c1, c2, scratch);
    }
    c1 /= N;
    c2 /= N;
    T* dxr = dx + (long long)row * N;
    const T* dhr = HAS_DH ? dh + (long long)row * N : nullptr;
    {
      int it = 0;
      for (int i = tid * W; i < N; i += nth * W, ++it) {
        typename VT::V dv = dyc[it];
        typename VT::V xv = xc[it];
        typename VT::V wv = VT::load(w + i);
        typename VT::V hv;
        if (HAS_DH) hv = VT::load(dhr + i);
        typename VT::V ov;
#pragma unroll
        for (int k = 0; k < W; ++k) {
          float d = VT::get(dv, k);
          float xh = (VT::get(xv, k) - mu) * rs;
          float wdy = VT::get(wv, k) * d;
          float g = (wdy - (xh * c1 + c2)) * rs;
          if (HAS_DH) g += VT::get(hv, k);
          VT::set(ov, k, g);
        }
        VT::store(dxr + i, ov);
      }
    }
  }
  // stripe write: block-owned rows of pdw/pdb
  if (STRIPES) {
    float* sdw = pdw + (long long)blockIdx.x * N;
    float* sdb = pdb + (long long)blockIdx.x * N;
    int it = 0;
    for (int i = tid * W; i < N; i += nth * W, ++it) {
#pragma unroll
      for (int k = 0; k < W; ++k) {
        sdw[i + k] = accdw[it][k];
        sdb[i + k] = accdb[it][k];
      }
    }
  }
}

// Streaming dw/db stripe accumulation (no barriers: the per-column sums
// need no cross-thread reduction). Block g chains rows g, g+G, ... exactly
// like the fused form, but with nothing serializing the row loop it runs
// at the HBM read floor.
template <typename T, int MAXITER>
__global__ void ln_dwdb_accum_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     float* __restrict__ pdw,
                                     float* __restrict__ pdb, int M, int N) {
  using VT = VecTraits<T>;
  constexpr int W = VT::W;
  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  float accdw[MAXITER][W];
  float accdb[MAXITER][W];
#pragma unroll
  for (int it = 0; it < MAXITER; ++it)
#pragma unroll
    for (int k = 0; k < W; ++k) accdw[it][k] = accdb[it][k] = 0.f;
  for (int row = blockIdx.x; row < M; row += gridDim.x) {
    const T* dyr = dy + (long long)row * N;
    const T* xr = x + (long long)row * N;
    const float mu = mean[row];
    const float rs = rstd[row];
    int it = 0;
    for (int i = tid * W; i < N; i += nth * W, ++it) {
      typename VT::V dv = VT::load(dyr + i);
      typename VT::V xv = VT::load(xr + i);
#pragma unroll
      for (int k = 0; k < W; ++k) {
        float d = VT::get(dv, k);
        accdw[it][k] += d * (VT::get(xv, k) - mu) * rs;
        accdb[it][k] += d;
      }
    }
  }
  float* sdw = pdw + (long long)blockIdx.x * N;
  float* sdb = pdb + (long long)blockIdx.x * N;
  int it = 0;
  for (int i = tid * W; i < N; i += nth * W, ++it) {
#pragma unroll
    for (int k = 0; k < W; ++k) {
      sdw[i + k] = accdw[it][k];
      sdb[i + k] = accdb[it][k];
    }
  }
}

// ---- wide-row fallback ----------------------------------------------------
// Rows beyond the register-cache bound (N > 2*block*W) use a two-pass
// grid-strided form that re-reads the row from global (L2-resident at
// practical widths). Correctness fallback for non-GPT widths — the narrow
// register-cached kernels above stay the hot path (VERDICT r1 weak #6:
// the old launcher returned hipErrorInvalidValue past N=4096 bf16).
template <typename T, bool HAS_RES>
__global__ void ln_fwd_wide_kernel(const T* __restrict__ x,
                                   const T* __restrict__ res, T* __restrict__ h,
                                   const T* __restrict__ w, const T* __restrict__ b,
                                   T* __restrict__ y, float* __restrict__ mean,
                                   float* __restrict__ rstd, int M, int N,
                                   float eps) {
  using VT = VecTraits<T>;
  constexpr int W = VT::W;
  __shared__ float scratch[2 * 1024 / WAVE];
  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  for (int row = blockIdx.x; row < M; row += gridDim.x) {
    const T* xr = x + (long long)row * N;
    const T* rr = HAS_RES ? res + (long long)row * N : nullptr;
    T* hr = HAS_RES ? h + (long long)row * N : nullptr;
    float s = 0.f, ss = 0.f;
    for (int i = tid * W; i < N; i += nth * W) {
      typename VT::V v = VT::load(xr + i);
      if (HAS_RES) {
        typename VT::V rv = VT::load(rr + i);
        typename VT::V hv;
#pragma unroll
        for (int k = 0; k < W; ++k)
          VT::set(hv, k, VT::get(v, k) + VT::get(rv, k));
        VT::store(hr + i, hv);
        v = hv;
      }
#pragma unroll
      for (int k = 0; k < W; ++k) {
        float f = VT::get(v, k);
        s += f;
        ss += f * f;
      }
    }
    block_sum2(s, ss, scratch);
    const float mu = s / N;
    const float var = fmaxf(ss / N - mu * mu, 0.0f);
    const float rs = rsqrtf(var + eps);
    if (tid == 0) {
      mean[row] = mu;
      rstd[row] = rs;
    }
    // pass 2: re-read the (already residual-fused) row
    const T* src = HAS_RES ? hr : xr;
    T* yr = y + (long long)row * N;
    for (int i = tid * W; i < N; i += nth * W) {
      typename VT::V xv = VT::load(src + i);
      typename VT::V wv = VT::load(w + i);
      typename VT::V bv = VT::load(b + i);
      typename VT::V ov;
#pragma unroll
      for (int k = 0; k < W; ++k) {
        float xh = (VT::get(xv, k) - mu) * rs;
        VT::set(ov, k, xh * VT::get(wv, k) + VT::get(bv, k));
      }
      VT::store(yr + i, ov);
    }
  }
}

// Wide backward: stripe partials accumulate straight into the block-owned
// global stripe rows (read-modify-write, no atomics needed) after a
// zero-init sweep — registers cannot hold a whole wide row.
template <typename T, bool HAS_DH>
__global__ void ln_bwd_dx_wide_kernel(const T* __restrict__ dy,
                                      const T* __restrict__ dh,
                                      const T* __restrict__ x,
                                      const T* __restrict__ w,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ rstd,
                                      T* __restrict__ dx, float* __restrict__ pdw,
                                      float* __restrict__ pdb, int M, int N) {
  using VT = VecTraits<T>;
  constexpr int W = VT::W;
  __shared__ float scratch[2 * 1024 / WAVE];
  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  float* sdw = pdw + (long long)blockIdx.x * N;
  float* sdb = pdb + (long long)blockIdx.x * N;
  for (int i = tid; i < N; i += nth) {
    sdw[i] = 0.f;
    sdb[i] = 0.f;
  }
  for (int row = blockIdx.x; row < M; row += gridDim.x) {
    const T* dyr = dy + (long long)row * N;
    const T* xr = x + (long long)row * N;
    const float mu = mean[row];
    const float rs = rstd[row];
    float c1 = 0.f, c2 = 0.f;
    for (int i = tid * W; i < N; i += nth * W) {
      typename VT::V dv = VT::load(dyr + i);
      typename VT::V xv = VT::load(xr + i);
      typename VT::V wv = VT::load(w + i);
#pragma unroll
      for (int k = 0; k < W; ++k) {
        float d = VT::get(dv, k);
        float xh = (VT::get(xv, k) - mu) * rs;
        float wdy = VT::get(wv, k) * d;
        c1 += xh * wdy;
        c2 += wdy;
      }
    }
    block_sum2(c1, c2, scratch);
    c1 /= N;
    c2 /= N;
    T* dxr = dx + (long long)row * N;
    const T* dhr = HAS_DH ? dh + (long long)row * N : nullptr;
    for (int i = tid * W; i < N; i += nth * W) {
      typename VT::V dv = VT::load(dyr + i);
      typename VT::V xv = VT::load(xr + i);
      typename VT::V wv = VT::load(w + i);
      typename VT::V hv;
      if (HAS_DH) hv = VT::load(dhr + i);
      typename VT::V ov;
#pragma unroll
      for (int k = 0; k < W; ++k) {
        float d = VT::get(dv, k);
        float xh = (VT::get(xv, k) - mu) * rs;
        float wdy = VT::get(wv, k) * d;
        float g = (wdy - (xh * c1 + c2)) * rs;
        if (HAS_DH) g += VT::get(hv, k);
        VT::set(ov, k, g);
        sdw[i + k] += d * xh;
        sdb[i + k] += d;
      }
      VT::store(dxr + i, ov);
    }
  }
}

// Column-reduce the stripe buffers -> dw[N], db[N] (fp32 out; caller casts).
// One wave per column (4 columns per block): lanes stride the G stripes,
// wave-reduce, lane 0 writes. No atomics, no zero-init launch (an earlier
// 2-D atomicAdd version needed two at::zeros fills per call — ~100 fill
// launches per step; and the first 1-D thread-per-column version was the
// single worst kernel in the bench profile at N/256 workgroups).
__global__ void ln_bwd_dwdb_kernel(const float* __restrict__ pdw,
                                   const float* __restrict__ pdb,
                                   float* __restrict__ dw, float* __restrict__ db,
                                   int G, int N) {
  const int c = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  if (c >= N) return;
  const int lane = threadIdx.x & (WAVE - 1);
  float sw = 0.f, sb = 0.f;
  for (int g = lane; g < G; g += WAVE) {
    sw += pdw[(long long)g * N + c];
    sb += pdb[(long long)g * N + c];
  }
  sw = wave_sum(sw);
  sb = wave_sum(sb);
  if (lane == 0) {
    dw[c] = sw;
    db[c] = sb;
  }
}

}  // namespace tdsa

using namespace tdsa;

extern "C" {

// Block size matched to the row width: at 256 threads and N=1024 bf16 only
// 128 lanes have work (tid*W < N) — half the block idled (profiled 2x).
static int ln_block(int N, int W) {
  int need = (N + W - 1) / W;
  int blk = ((need + 63) / 64) * 64;
  if (blk > 256) blk = 256;
  if (blk < 64) blk = 64;
  return blk;
}

// res/h: optional fused residual (pass nullptr for the plain form).
hipError_t tdsa_ln_fwd(const void* x, const void* res, void* h, const void* w,
                       const void* b, void* y, float* mean, float* rstd,
                       int M, int N, float eps, int is_bf16,
                       hipStream_t stream) {
  const int block = ln_block(N, is_bf16 ? 8 : 4);
  // one row per block up to the cap: chaining rows serializes the per-row
  // reduction barriers; resident blocks overlap freely instead
  int cap = 32768;
  if (const char* v = getenv("TDSA_LN_GRID")) cap = atoi(v);
  const int grid = (M < cap) ? M : cap;
  // one-wave rows (no barriers): measured 2-3x SLOWER than the 2-wave
  // block form (64-lane blocks lose memory-level parallelism; the barrier
  // was never the bottleneck) — kept behind TDSA_LN_WAVE=1 as a record
  int wave_ok = 0;
  if (const char* v = getenv("TDSA_LN_WAVE")) wave_ok = atoi(v);
#define LN_FWD(T, HASR)                                                       \
  hipLaunchKernelGGL((ln_fwd_kernel<T, HASR>), dim3(grid), dim3(block), 0,    \
                     stream, (const T*)x, (const T*)res, (T*)h, (const T*)w,  \
                     (const T*)b, (T*)y, mean, rstd, M, N, eps)
#define LN_FWD_OW(T, HASR)                                                    \
  hipLaunchKernelGGL((ln_fwd_kernel<T, HASR, 4, true>), dim3(grid),           \
                     dim3(WAVE), 0, stream, (const T*)x, (const T*)res,       \
                     (T*)h, (const T*)w, (const T*)b, (T*)y, mean, rstd, M,   \
                     N, eps)
#define LN_FWD_WIDE(T, HASR)                                                  \
  hipLaunchKernelGGL((ln_fwd_wide_kernel<T, HASR>), dim3(grid), dim3(1024),   \
                     0, stream, (const T*)x, (const T*)res, (T*)h,            \
                     (const T*)w, (const T*)b, (T*)y, mean, rstd, M, N, eps)
  if (is_bf16) {
    if (N % 8) return hipErrorInvalidValue;  // 16B row-base alignment
    if (wave_ok && N <= 4 * WAVE * 8) {
      if (res) LN_FWD_OW(bf16, true); else LN_FWD_OW(bf16, false);
    } else if (N > 2 * block * 8) {
      if (res) LN_FWD_WIDE(bf16, true); else LN_FWD_WIDE(bf16, false);
    } else {
      if (res) LN_FWD(bf16, true); else LN_FWD(bf16, false);
    }
  } else {
    if (N % 4) return hipErrorInvalidValue;
    if (wave_ok && N <= 4 * WAVE * 4) {
      if (res) LN_FWD_OW(float, true); else LN_FWD_OW(float, false);
    } else if (N > 2 * block * 4) {
      if (res) LN_FWD_WIDE(float, true); else LN_FWD_WIDE(float, false);
    } else {
      if (res) LN_FWD(float, true); else LN_FWD(float, false);
    }
  }
#undef LN_FWD
#undef LN_FWD_OW
#undef LN_FWD_WIDE
  return hipGetLastError();
}

// G (stripe count) is chosen here and reported to the caller so it can size
// pdw/pdb; call with pdw==nullptr to query G.
int tdsa_ln_bwd_dx_stripes(int M) {
  // within-box sweep: 2048 stripes beat 4096 (dwdb reduce pays per stripe)
  int cap = 2048;
  if (const char* v = getenv("TDSA_LN_STRIPES")) cap = atoi(v);
  int g = M < cap ? M : cap;
  return g < 1 ? 1 : g;
}

// dh: optional residual-stream gradient added into dx (nullptr for plain).
// Default is the SPLIT scheme (dx at full grid + streaming stripe accum;
// see ln_bwd_dx_kernel comment); TDSA_LN_SPLIT=0 restores the fused form.
hipError_t tdsa_ln_bwd_dx(const void* dy, const void* dh, const void* x,
                          const void* w, const float* mean, const float* rstd,
                          void* dx, float* pdw, float* pdb, int M, int N,
                          int is_bf16, hipStream_t stream) {
  const int block = ln_block(N, is_bf16 ? 8 : 4);
  const int grid = tdsa_ln_bwd_dx_stripes(M);
  int split = 1;
  if (const char* v = getenv("TDSA_LN_SPLIT")) split = atoi(v);
  int dx_cap = 32768;
  if (const char* v = getenv("TDSA_LN_GRID")) dx_cap = atoi(v);
  const int dx_grid = split ? ((M < dx_cap) ? M : dx_cap) : grid;
  int wave_ok = 0;  // measured worse (see tdsa_ln_fwd comment)
  if (const char* v = getenv("TDSA_LN_WAVE")) wave_ok = atoi(v);
  const int ow = wave_ok && N <= 4 * WAVE * (is_bf16 ? 8 : 4);
#define LN_BWD(T, HASD)                                                        \
  do {                                                                         \
    if (split && ow) {                                                         \
      hipLaunchKernelGGL((ln_bwd_dx_kernel<T, 4, HASD, false, true>),          \
                         dim3(dx_grid), dim3(WAVE), 0, stream, (const T*)dy,   \
                         (const T*)dh, (const T*)x, (const T*)w, mean, rstd,   \
                         (T*)dx, pdw, pdb, M, N);                              \
      hipLaunchKernelGGL((ln_dwdb_accum_kernel<T, 2>), dim3(grid),             \
                         dim3(block), 0, stream, (const T*)dy, (const T*)x,    \
                         mean, rstd, pdw, pdb, M, N);                          \
    } else if (split) {                                                        \
      hipLaunchKernelGGL((ln_bwd_dx_kernel<T, 2, HASD, false>), dim3(dx_grid), \
                         dim3(block), 0, stream, (const T*)dy, (const T*)dh,   \
                         (const T*)x, (const T*)w, mean, rstd, (T*)dx, pdw,    \
                         pdb, M, N);                                           \
      hipLaunchKernelGGL((ln_dwdb_accum_kernel<T, 2>), dim3(grid),             \
                         dim3(block), 0, stream, (const T*)dy, (const T*)x,    \
                         mean, rstd, pdw, pdb, M, N);                          \
    } else {                                                                   \
      hipLaunchKernelGGL((ln_bwd_dx_kernel<T, 2, HASD, true>), dim3(grid),     \
                         dim3(block), 0, stream, (const T*)dy, (const T*)dh,   \
                         (const T*)x, (const T*)w, mean, rstd, (T*)dx, pdw,    \
                         pdb, M, N);                                           \
    }                                                                          \
  } while (0)
#define LN_BWD_WIDE(T, HASD)                                                  \
  hipLaunchKernelGGL((ln_bwd_dx_wide_kernel<T, HASD>), dim3(grid),            \
                     dim3(1024), 0, stream, (const T*)dy, (const T*)dh,       \
                     (const T*)x, (const T*)w, mean, rstd, (T*)dx, pdw, pdb,  \
                     M, N)
  if (is_bf16) {
    if (N % 8) return hipErrorInvalidValue;  // 16B row-base alignment
    if (N > 2 * block * 8) {
      if (dh) LN_BWD_WIDE(bf16, true); else LN_BWD_WIDE(bf16, false);
    } else {
      if (dh) LN_BWD(bf16, true); else LN_BWD(bf16, false);
    }
  } else {
    if (N % 4) return hipErrorInvalidValue;
    if (N > 2 * block * 4) {
      if (dh) LN_BWD_WIDE(float, true); else LN_BWD_WIDE(float, false);
    } else {
      if (dh) LN_BWD(float, true); else LN_BWD(float, false);
    }
  }
#undef LN_BWD
#undef LN_BWD_WIDE
  return hipGetLastError();
}

hipError_t tdsa_ln_bwd_dwdb(const float* pdw, const float* pdb, float* dw,
                            float* db, int G, int N, hipStream_t stream) {
  const int block = 256;
  const int cols_per_block = block / WAVE;
  const int grid = (N + cols_per_block - 1) / cols_per_block;
  hipLaunchKernelGGL(ln_bwd_dwdb_kernel, dim3(grid), dim3(block), 0, stream,
                     pdw, pdb, dw, db, G, N);
  return hipGetLastError();
}

}  // extern "C"
