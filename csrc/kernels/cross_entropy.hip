// Fused softmax cross-entropy CDNA4 kernels.
//
// The reference calls F.cross_entropy on (B*T, 50304) logits
// (/root/reference/example/model.py:156). Here the forward is ONE HBM pass
// per row: an online logsumexp (running max + rescaled sum, the same merge
// rule as flash attention) plus the picked-logit lookup; backward is one
// pass writing (softmax - onehot) * scale. fp32 accumulation throughout.
#include "common.h"

#include <cstdlib>

namespace tdsa {

// merge two (m, s) logsumexp states (raw v_exp: exp(-inf) == 0, and the
// branchless form keeps both scales well-defined for m == m2 == -inf)
DEV_INLINE void lse_merge(float& m, float& s, float m2, float s2) {
  float mn = fmaxf(m, m2);
  float a = (m == -INFINITY) ? 0.f
            : __builtin_amdgcn_exp2f(1.4426950408889634f * (m - mn));
  float b = (m2 == -INFINITY) ? 0.f
            : __builtin_amdgcn_exp2f(1.4426950408889634f * (m2 - mn));
  s = s * a + s2 * b;
  m = mn;
}

template <typename T, int W>
struct RowVec;
template <> struct RowVec<float, 4> {
  typedef float4v V;
  static DEV_INLINE V load(const float* p) { return *reinterpret_cast<const V*>(p); }
  static DEV_INLINE float get(V v, int i) { return v[i]; }
  static DEV_INLINE void set(V& v, int i, float x) { v[i] = x; }
  static DEV_INLINE void store(float* p, V v) { *reinterpret_cast<V*>(p) = v; }
};
template <> struct RowVec<bf16, 8> {
  typedef short8v V;
  static DEV_INLINE V load(const bf16* p) { return load8(p); }
  static DEV_INLINE float get(V v, int i) { return bf_elem(v, i); }
  static DEV_INLINE void set(V& v, int i, float x) { v[i] = bf_pack(x); }
  static DEV_INLINE void store(bf16* p, V v) { store8(p, v); }
};

// One workgroup per row (grid-strided). Writes lse[row] (fp32) and
// atomically accumulates loss_sum (fp32 scalar) and n_valid (int32).
template <typename T, int W>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const long long* __restrict__ targets,
                              float* __restrict__ lse, float* __restrict__ loss_sum,
                              int* __restrict__ n_valid, long long R, int V,
                              long long ignore_index) {
  using RV = RowVec<T, W>;
  __shared__ float sm[1024 / WAVE];
  __shared__ float ssum[1024 / WAVE];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int nw = blockDim.x / WAVE;
  // block-local loss/count accumulators: one atomicAdd per BLOCK at the end
  // (a per-row atomicAdd on the single loss scalar serialized 32k blocks —
  // measured 4x the kernel's HBM floor)
  float loss_acc = 0.f;
  int valid_acc = 0;
  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const T* lr = logits + row * V;
    float m = -INFINITY, s = 0.f;
    // per-CHUNK online merge: a per-element data-dependent branch diverges
    // and serializes; instead reduce the chunk max, sum exps against it,
    // and merge (m, s) once per chunk.
    for (int i = tid * W; i + W <= V; i += blockDim.x * W) {
      typename RV::V v = RV::load(lr + i);
      float cm = RV::get(v, 0);
#pragma unroll
      for (int k = 1; k < W; ++k) cm = fmaxf(cm, RV::get(v, k));
      float cs = 0.f;
#pragma unroll
      for (int k = 0; k < W; ++k)
        cs += __builtin_amdgcn_exp2f(1.4426950408889634f
                                     * (RV::get(v, k) - cm));
      lse_merge(m, s, cm, cs);
    }
    const int rem0 = (V / W) * W;
    for (int j = rem0 + tid; j < V; j += blockDim.x) {
      float f = (float)lr[j];
      if (f > m) {
        s = s * __expf(m - f) + 1.0f;
        m = f;
      } else {
        s += __expf(f - m);
      }
    }
    // wave merge
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float m2 = __shfl_xor(m, off, WAVE);
      float s2 = __shfl_xor(s, off, WAVE);
      lse_merge(m, s, m2, s2);
    }
    if (lane == 0) { sm[wid] = m; ssum[wid] = s; }
    __syncthreads();
    if (wid == 0) {
      float mm = (lane < nw) ? sm[lane] : -INFINITY;
      float ss2 = (lane < nw) ? ssum[lane] : 0.f;
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        float m2 = __shfl_xor(mm, off, WAVE);
        float s2 = __shfl_xor(ss2, off, WAVE);
        lse_merge(mm, ss2, m2, s2);
      }
      if (lane == 0) {
        float l = mm + __logf(ss2);
        lse[row] = l;
        const long long t = targets[row];
        if (t != ignore_index) {
          float picked = (float)lr[t];
          loss_acc += l - picked;
          valid_acc += 1;
        }
      }
    }
    __syncthreads();
  }
  if (tid == 0 && valid_acc > 0) {
    atomicAdd(loss_sum, loss_acc);
    atomicAdd(n_valid, valid_acc);
  }
}

// dlogits = valid ? (exp(logit - lse) - onehot) * scale : 0
template <typename T, int W>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const long long* __restrict__ targets,
                              const float* __restrict__ lse, T* __restrict__ dlogits,
                              long long R, int V, float scale,
                              long long ignore_index) {
  using RV = RowVec<T, W>;
  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const T* lr = logits + row * V;
    T* dr = dlogits + row * V;
    const long long t = targets[row];
    const bool valid = (t != ignore_index);
    const float l = lse[row];
    for (int i = threadIdx.x * W; i + W <= V; i += blockDim.x * W) {
      typename RV::V v = RV::load(lr + i);
      typename RV::V out;
#pragma unroll
      for (int k = 0; k < W; ++k) {
        float g = 0.f;
        if (valid) {
          g = __expf(RV::get(v, k) - l);
          if ((long long)(i + k) == t) g -= 1.0f;
          g *= scale;
        }
        RV::set(out, k, g);
      }
      RV::store(dr + i, out);
    }
    // tail
    int rem0 = (V / W) * W;
    for (int j = rem0 + threadIdx.x; j < V; j += blockDim.x) {
      float g = 0.f;
      if (valid) {
        g = __expf((float)lr[j] - l);
        if ((long long)j == t) g -= 1.0f;
        g *= scale;
      }
      dr[j] = (T)g;
    }
  }
}

}  // namespace tdsa

using namespace tdsa;

extern "C" {

hipError_t tdsa_ce_fwd(const void* logits, const long long* targets, float* lse,
                       float* loss_sum, int* n_valid, long long R, int V,
                       long long ignore_index, int is_bf16, hipStream_t stream) {
  const int block = 256;
  // Row bases logits+row*V must stay 16B-aligned for the vector loads:
  // reject V % W != 0 (mirrors the layernorm launcher guard; the python op
  // routes such shapes to the composite path instead).
  if (V % (is_bf16 ? 8 : 4)) return hipErrorInvalidValue;
  long long cap = 32768;
  if (const char* v = getenv("TDSA_CE_GRID")) cap = atoll(v);
  const int grid = (int)((R < cap) ? R : cap);
  if (is_bf16)
    hipLaunchKernelGGL((ce_fwd_kernel<bf16, 8>), dim3(grid), dim3(block), 0,
                       stream, (const bf16*)logits, targets, lse, loss_sum,
                       n_valid, R, V, ignore_index);
  else
    hipLaunchKernelGGL((ce_fwd_kernel<float, 4>), dim3(grid), dim3(block), 0,
                       stream, (const float*)logits, targets, lse, loss_sum,
                       n_valid, R, V, ignore_index);
  return hipGetLastError();
}

hipError_t tdsa_ce_bwd(const void* logits, const long long* targets,
                       const float* lse, void* dlogits, long long R, int V,
                       float scale, long long ignore_index, int is_bf16,
                       hipStream_t stream) {
  const int block = 256;
  if (V % (is_bf16 ? 8 : 4)) return hipErrorInvalidValue;  // see tdsa_ce_fwd
  long long cap = 32768;
  if (const char* v = getenv("TDSA_CE_GRID")) cap = atoll(v);
  const int grid = (int)((R < cap) ? R : cap);
  if (is_bf16)
    hipLaunchKernelGGL((ce_bwd_kernel<bf16, 8>), dim3(grid), dim3(block), 0,
                       stream, (const bf16*)logits, targets, lse, (bf16*)dlogits,
                       R, V, scale, ignore_index);
  else
    hipLaunchKernelGGL((ce_bwd_kernel<float, 4>), dim3(grid), dim3(block), 0,
                       stream, (const float*)logits, targets, lse,
                       (float*)dlogits, R, V, scale, ignore_index);
  return hipGetLastError();
}

}  // extern "C"
