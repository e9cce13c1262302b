// Embedding gather / scatter-add CDNA4 kernels.
//
// Parity with /root/reference/tiny_deepspeed/core/module/ops/embedding.py:
// forward = row gather (index_select :57), backward = scatter-add into the
// (vocab, E) table (:60-65). MI355X design: vectorized 16B-per-lane gather;
// backward accumulates into an fp32 buffer with device-scope atomicAdd
// (bf16 atomics would lose small contributions).
#include "common.h"

namespace tdsa {

// vector type of width W over T
template <typename T, int W> struct VecOf;
template <> struct VecOf<float, 4> { typedef float4v type; };
template <> struct VecOf<bf16, 8> { typedef short8v type; };

template <typename T>
__global__ void emb_fwd_scalar(const T* __restrict__ weight,
                               const long long* __restrict__ idx,
                               T* __restrict__ out, long long R, int D) {
  const long long total = R * D;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = global_tid(); i < total; i += stride) {
    const long long r = i / D;
    const int c = (int)(i % D);
    out[i] = weight[idx[r] * (long long)D + c];
  }
}

template <typename T, int W>
__global__ void emb_fwd_vec(const T* __restrict__ weight,
                            const long long* __restrict__ idx,
                            T* __restrict__ out, long long R, int D) {
  typedef typename VecOf<T, W>::type V;
  const int chunks = D / W;
  const long long total = R * chunks;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = global_tid(); i < total; i += stride) {
    const long long r = i / chunks;
    const int c = (int)(i % chunks) * W;
    *reinterpret_cast<V*>(out + r * (long long)D + c) =
        *reinterpret_cast<const V*>(weight + idx[r] * (long long)D + c);
  }
}

template <typename T>
__global__ void emb_bwd_kernel(const T* __restrict__ dy,
                               const long long* __restrict__ idx,
                               float* __restrict__ dw32, long long R, int D,
                               long long padding_idx) {
  const long long total = R * D;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = global_tid(); i < total; i += stride) {
    const long long r = i / D;
    const int c = (int)(i % D);
    const long long t = idx[r];
    if (t == padding_idx) continue;
    atomicAdd(&dw32[t * (long long)D + c], (float)dy[i]);
  }
}

}  // namespace tdsa

using namespace tdsa;

extern "C" {

hipError_t tdsa_embedding_fwd(const void* weight, const long long* idx, void* out,
                              long long R, int D, int is_bf16, hipStream_t stream) {
  const int block = 256;
  if (is_bf16 && D % 8 == 0) {
    long long work = R * (D / 8);
    hipLaunchKernelGGL((emb_fwd_vec<bf16, 8>), dim3(ln_grid(work, block)),
                       dim3(block), 0, stream, (const bf16*)weight, idx,
                       (bf16*)out, R, D);
  } else if (!is_bf16 && D % 4 == 0) {
    long long work = R * (D / 4);
    hipLaunchKernelGGL((emb_fwd_vec<float, 4>), dim3(ln_grid(work, block)),
                       dim3(block), 0, stream, (const float*)weight, idx,
                       (float*)out, R, D);
  } else if (is_bf16) {
    hipLaunchKernelGGL(emb_fwd_scalar<bf16>, dim3(ln_grid(R * D, block)),
                       dim3(block), 0, stream, (const bf16*)weight, idx,
                       (bf16*)out, R, D);
  } else {
    hipLaunchKernelGGL(emb_fwd_scalar<float>, dim3(ln_grid(R * D, block)),
                       dim3(block), 0, stream, (const float*)weight, idx,
                       (float*)out, R, D);
  }
  return hipGetLastError();
}

// dw32 must be zero-filled fp32 [V, D]; padding_idx = -1 for "none".
hipError_t tdsa_embedding_bwd(const void* dy, const long long* idx, float* dw32,
                              long long R, int D, long long padding_idx,
                              int is_bf16, hipStream_t stream) {
  const int block = 256;
  if (is_bf16)
    hipLaunchKernelGGL(emb_bwd_kernel<bf16>, dim3(ln_grid(R * D, block)),
                       dim3(block), 0, stream, (const bf16*)dy, idx, dw32, R, D,
                       padding_idx);
  else
    hipLaunchKernelGGL(emb_bwd_kernel<float>, dim3(ln_grid(R * D, block)),
                       dim3(block), 0, stream, (const float*)dy, idx, dw32, R, D,
                       padding_idx);
  return hipGetLastError();
}

}  // extern "C"
