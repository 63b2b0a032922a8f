// Fused updaters: the whole reference update chain in one kernel per param
// (SURVEY.md §2.3 optimizer kernels):
//   g' = clamp(g, -clip, clip) + l2 * w          (ClipElementWiseAbsoluteValue
//                                                 + coupled L2, Java:123-125)
//   Adam:    m = b1*m + (1-b1)*g'; v = b2*v + (1-b2)*g'^2
//            w -= lr * (m/(1-b1^t)) / (sqrt(v/(1-b2^t)) + eps)
//   RMSProp: v = d*v + (1-d)*g'^2;  w -= lr * g' / (sqrt(v) + eps)
// Master weights are fp32; the bf16 compute copy is written back in the
// same kernel (no separate cast pass over HBM).

#include "common.h"

template <typename GT>
DEV_INLINE float load_g(const GT* g, long i);
template <>
DEV_INLINE float load_g<unsigned short>(const unsigned short* g, long i) {
  return bf2f(g[i]);
}
template <>
DEV_INLINE float load_g<float>(const float* g, long i) { return g[i]; }

template <typename GT, bool BF16_PARAM>
__global__ void fused_adam_k(void* __restrict__ param,
                             const GT* __restrict__ grad,
                             float* __restrict__ master,
                             float* __restrict__ m, float* __restrict__ v,
                             long n, float lr, float b1, float b2, float eps,
                             float clip, float l2, float bc1, float bc2,
                             const int* __restrict__ t_dev) {
  if (t_dev != nullptr) {
    // hipGraph-replayable: step count lives on-device (host scalars are
    // frozen at capture), so bias correction is computed here
    float t = (float)*t_dev;
    bc1 = 1.f - __powf(b1, t);
    bc2 = 1.f - __powf(b2, t);
  }
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (long)gridDim.x * blockDim.x) {
    float w = BF16_PARAM ? master[i] : ((float*)param)[i];
    float g = load_g(grad, i);
    if (clip > 0.f) g = fminf(fmaxf(g, -clip), clip);
    g += l2 * w;
    float mi = b1 * m[i] + (1.f - b1) * g;
    float vi = b2 * v[i] + (1.f - b2) * g * g;
    m[i] = mi;
    v[i] = vi;
    w -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
    if (BF16_PARAM) {
      master[i] = w;
      ((unsigned short*)param)[i] = f2bf(w);
    } else {
      ((float*)param)[i] = w;
    }
  }
}

template <typename GT, bool BF16_PARAM>
__global__ void fused_rmsprop_k(void* __restrict__ param,
                                const GT* __restrict__ grad,
                                float* __restrict__ master,
                                float* __restrict__ v, long n, float lr,
                                float decay, float eps, float clip, float l2) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (long)gridDim.x * blockDim.x) {
    float w = BF16_PARAM ? master[i] : ((float*)param)[i];
    float g = load_g(grad, i);
    if (clip > 0.f) g = fminf(fmaxf(g, -clip), clip);
    g += l2 * w;
    float vi = decay * v[i] + (1.f - decay) * g * g;
    v[i] = vi;
    w -= lr * g / (sqrtf(vi) + eps);
    if (BF16_PARAM) {
      master[i] = w;
      ((unsigned short*)param)[i] = f2bf(w);
    } else {
      ((float*)param)[i] = w;
    }
  }
}

extern "C" {

void launch_fused_adam(void* param, const void* grad, float* master, float* m,
                       float* v, long n, int grad_is_bf16, int param_is_bf16,
                       float lr, float b1, float b2, float eps, float clip,
                       float l2, int t, const int* t_dev, hipStream_t s) {
  float bc1 = 1.f - powf(b1, (float)t);
  float bc2 = 1.f - powf(b2, (float)t);
  int grid = (int)min((long)2048, (n + 255) / 256 + 1);
  if (grad_is_bf16 && param_is_bf16)
    hipLaunchKernelGGL((fused_adam_k<unsigned short, true>), dim3(grid),
                       dim3(256), 0, s, param, (const unsigned short*)grad,
                       master, m, v, n, lr, b1, b2, eps, clip, l2, bc1, bc2, t_dev);
  else if (!grad_is_bf16 && param_is_bf16)
    hipLaunchKernelGGL((fused_adam_k<float, true>), dim3(grid), dim3(256), 0,
                       s, param, (const float*)grad, master, m, v, n, lr, b1,
                       b2, eps, clip, l2, bc1, bc2, t_dev);
  else if (grad_is_bf16)
    hipLaunchKernelGGL((fused_adam_k<unsigned short, false>), dim3(grid),
                       dim3(256), 0, s, param, (const unsigned short*)grad,
                       master, m, v, n, lr, b1, b2, eps, clip, l2, bc1, bc2, t_dev);
  else
    hipLaunchKernelGGL((fused_adam_k<float, false>), dim3(grid), dim3(256), 0,
                       s, param, (const float*)grad, master, m, v, n, lr, b1,
                       b2, eps, clip, l2, bc1, bc2, t_dev);
}

void launch_fused_rmsprop(void* param, const void* grad, float* master,
                          float* v, long n, int grad_is_bf16,
                          int param_is_bf16, float lr, float decay, float eps,
                          float clip, float l2, hipStream_t s) {
  int grid = (int)min((long)2048, (n + 255) / 256 + 1);
  if (grad_is_bf16 && param_is_bf16)
    hipLaunchKernelGGL((fused_rmsprop_k<unsigned short, true>), dim3(grid),
                       dim3(256), 0, s, param, (const unsigned short*)grad,
                       master, v, n, lr, decay, eps, clip, l2);
  else if (!grad_is_bf16 && param_is_bf16)
    hipLaunchKernelGGL((fused_rmsprop_k<float, true>), dim3(grid), dim3(256),
                       0, s, param, (const float*)grad, master, v, n, lr,
                       decay, eps, clip, l2);
  else if (grad_is_bf16)
    hipLaunchKernelGGL((fused_rmsprop_k<unsigned short, false>), dim3(grid),
                       dim3(256), 0, s, param, (const unsigned short*)grad,
                       master, v, n, lr, decay, eps, clip, l2);
  else
    hipLaunchKernelGGL((fused_rmsprop_k<float, false>), dim3(grid), dim3(256),
                       0, s, param, (const float*)grad, master, v, n, lr,
                       decay, eps, clip, l2);
}

}  // extern "C"
