// Fused optimizer updates (SGD+momentum, AdamW), exact torch.optim semantics
// (reference optimizer sites: src/train/VGG16.py:62 SGD(lr, momentum);
// src/train/BERT.py:69 / KWT.py:62 AdamW(lr, weight_decay)).
// The C++ wrapper loops tensors and launches one elementwise kernel per tensor;
// per-microbatch step capture into a hipGraph absorbs the launch overhead.
#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace slk {

// zero_after: clears the gradient after applying it, replacing the per-tensor
// zero_grad fill kernels (autograd then accumulates the next microbatch's
// gradients into already-zeroed buffers).
__global__ void sgd_kernel(float* __restrict__ p, float* __restrict__ g,
                           float* __restrict__ buf, long n, float lr, float momentum,
                           float weight_decay, bool first, bool zero_after) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float grad = g[i];
    if (zero_after) g[i] = 0.f;
    if (weight_decay != 0.f) grad += weight_decay * p[i];
    float b;
    if (momentum != 0.f) {
      b = first ? grad : momentum * buf[i] + grad;  // torch dampening=0
      buf[i] = b;
    } else {
      b = grad;
    }
    p[i] -= lr * b;
  }
}

__global__ void adamw_kernel(float* __restrict__ p, float* __restrict__ g,
                             float* __restrict__ m, float* __restrict__ v, long n,
                             float lr, float beta1, float beta2, float eps,
                             float weight_decay, float bc1, float bc2,
                             bool zero_after) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    // decoupled weight decay (AdamW): p *= (1 - lr*wd)
    float pv = p[i] * (1.f - lr * weight_decay);
    const float grad = g[i];
    if (zero_after) g[i] = 0.f;
    const float mi = beta1 * m[i] + (1.f - beta1) * grad;
    const float vi = beta2 * v[i] + (1.f - beta2) * grad * grad;
    m[i] = mi;
    v[i] = vi;
    const float mhat = mi / bc1;
    const float vhat = vi / bc2;
    p[i] = pv - lr * mhat / (sqrtf(vhat) + eps);
  }
}

static inline int opt_grid(long n) {
  return (int)std::min<long>((n + 255) / 256, 2048);
}

void sgd_step(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
              std::vector<at::Tensor> bufs, double lr, double momentum,
              double weight_decay, bool first, bool zero_after) {
  auto stream = c10::hip::getCurrentHIPStream().stream();
  for (size_t i = 0; i < params.size(); ++i) {
    const long n = params[i].numel();
    hipLaunchKernelGGL(sgd_kernel, dim3(opt_grid(n)), dim3(256), 0, stream,
                       params[i].data_ptr<float>(), grads[i].data_ptr<float>(),
                       bufs[i].data_ptr<float>(), n, (float)lr, (float)momentum,
                       (float)weight_decay, first, zero_after);
  }
}

void adamw_step(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs, int64_t step,
                double lr, double beta1, double beta2, double eps,
                double weight_decay, bool zero_after) {
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  for (size_t i = 0; i < params.size(); ++i) {
    const long n = params[i].numel();
    hipLaunchKernelGGL(adamw_kernel, dim3(opt_grid(n)), dim3(256), 0, stream,
                       params[i].data_ptr<float>(), grads[i].data_ptr<float>(),
                       ms[i].data_ptr<float>(), vs[i].data_ptr<float>(), n, (float)lr,
                       (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                       bc1, bc2, zero_after);
  }
}

}  // namespace slk
