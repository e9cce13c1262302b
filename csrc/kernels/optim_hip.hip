#include "hip/hip_runtime.h"
// Fused optimizer-update CDNA4 kernels: AdamW and SGD(momentum).
//
// The reference's update math is 7+ separate torch kernel launches per
// parameter (/root/reference/tiny_deepspeed/core/optim/adamw.py:32-59,
// sgd.py:28-46). Here one kernel per parameter does the whole m/v/p update
// in a single HBM pass (fp32 state, fp32 master copy for bf16 params), with
// the reference's per-parameter step-count bug fixed: `step` is the global
// 1-based optimizer step.
#include "common.h"

namespace {

template <typename PT, typename GT>
__global__ void adamw_kernel(PT* __restrict__ param, const GT* __restrict__ grad,
                             float* __restrict__ m, float* __restrict__ v,
                             float* __restrict__ master, float* __restrict__ vmax,
                             int has_master, int amsgrad, float lr, float b1,
                             float b2, float eps, float wd, float inv_bc1,
                             float inv_bc2, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = global_tid(); i < n; i += stride) {
    float g = (float)grad[i];
    float p = has_master ? master[i] : (float)param[i];
    p *= (1.0f - lr * wd);
    float mi = m[i] = m[i] * b1 + (1.0f - b1) * g;
    float vi = v[i] = v[i] * b2 + (1.0f - b2) * g * g;
    float vhat;
    if (amsgrad) {
      float mx = fmaxf(vmax[i], vi);
      vmax[i] = mx;
      vhat = mx * inv_bc2;
    } else {
      vhat = vi * inv_bc2;
    }
    p -= lr * inv_bc1 * mi / (sqrtf(vhat) + eps);
    if (has_master) master[i] = p;
    param[i] = (PT)p;
  }
}

template <typename PT, typename GT>
__global__ void sgd_kernel(PT* __restrict__ param, const GT* __restrict__ grad,
                           float* __restrict__ buf, float* __restrict__ master,
                           int has_buf, int has_master, float lr, float momentum,
                           float dampening, float wd, int nesterov, int maximize,
                           int first_step, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = global_tid(); i < n; i += stride) {
    float g = (float)grad[i];
    if (maximize) g = -g;
    float p = has_master ? master[i] : (float)param[i];
    if (wd != 0.0f) g += wd * p;
    if (has_buf) {
      float b = first_step ? g : buf[i] * momentum + (1.0f - dampening) * g;
      buf[i] = b;
      g = nesterov ? (g + momentum * b) : b;
    }
    p -= lr * g;
    if (has_master) master[i] = p;
    param[i] = (PT)p;
  }
}

}  // namespace

extern "C" {

hipError_t tdsa_adamw_step(void* param, const void* grad, float* m, float* v,
                           float* master, float* vmax, int has_master,
                           int amsgrad, float lr, float b1, float b2, float eps,
                           float wd, long long step, long long n,
                           int param_bf16, int grad_bf16, hipStream_t stream) {
  const int block = 256;
  const int grid = ln_grid(n, block);
  const float inv_bc1 = 1.0f / (1.0f - powf(b1, (float)step));
  const float inv_bc2 = 1.0f / (1.0f - powf(b2, (float)step));
#define LAUNCH_ADAMW(PT, GT)                                                  \
  hipLaunchKernelGGL((adamw_kernel<PT, GT>), dim3(grid), dim3(block), 0,      \
                     stream, (PT*)param, (const GT*)grad, m, v, master, vmax, \
                     has_master, amsgrad, lr, b1, b2, eps, wd, inv_bc1,       \
                     inv_bc2, n)
  if (param_bf16 && grad_bf16) LAUNCH_ADAMW(bf16, bf16);
  else if (param_bf16) LAUNCH_ADAMW(bf16, float);
  else if (grad_bf16) LAUNCH_ADAMW(float, bf16);
  else LAUNCH_ADAMW(float, float);
#undef LAUNCH_ADAMW
  return hipGetLastError();
}

hipError_t tdsa_sgd_step(void* param, const void* grad, float* buf, float* master,
                         int has_buf, int has_master, float lr, float momentum,
                         float dampening, float wd, int nesterov, int maximize,
                         int first_step, long long n, int param_bf16,
                         int grad_bf16, hipStream_t stream) {
  const int block = 256;
  const int grid = ln_grid(n, block);
#define LAUNCH_SGD(PT, GT)                                                    \
  hipLaunchKernelGGL((sgd_kernel<PT, GT>), dim3(grid), dim3(block), 0, stream, \
                     (PT*)param, (const GT*)grad, buf, master, has_buf,        \
                     has_master, lr, momentum, dampening, wd, nesterov,        \
                     maximize, first_step, n)
  if (param_bf16 && grad_bf16) LAUNCH_SGD(bf16, bf16);
  else if (param_bf16) LAUNCH_SGD(bf16, float);
  else if (grad_bf16) LAUNCH_SGD(float, bf16);
  else LAUNCH_SGD(float, float);
#undef LAUNCH_SGD
  return hipGetLastError();
}

}  // extern "C"
