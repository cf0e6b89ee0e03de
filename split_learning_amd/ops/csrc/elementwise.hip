// Elementwise / pooling / dropout / embedding kernels (fp32, grid-stride).
// Memory-bound ops: float4-vectorized main loop with scalar tail (CDNA guide
// Appendix B: elementwise target is the HBM ceiling).
#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace slk {

static inline int ew_grid(long n, int threads = 256, int vec = 4) {
  long blocks = (n + (long)threads * vec - 1) / ((long)threads * vec);
  return (int)std::min<long>(blocks, 4096);
}

// ---------------- ReLU ----------------
__global__ void relu_fwd_kernel(const float4* __restrict__ x4, float4* __restrict__ y4,
                                const float* __restrict__ x, float* __restrict__ y,
                                long n4, long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    float4 v = x4[i];
    v.x = fmaxf(v.x, 0.f); v.y = fmaxf(v.y, 0.f);
    v.z = fmaxf(v.z, 0.f); v.w = fmaxf(v.w, 0.f);
    y4[i] = v;
  }
  for (long i = n4 * 4 + (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    y[i] = fmaxf(x[i], 0.f);
}

__global__ void relu_bwd_kernel(const float* __restrict__ gy,
                                const float* __restrict__ y, float* __restrict__ gx,
                                long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    gx[i] = y[i] > 0.f ? gy[i] : 0.f;
}

at::Tensor relu_fwd(const at::Tensor& x) {
  auto y = at::empty_like(x);
  const long n = x.numel(), n4 = n / 4;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(relu_fwd_kernel, dim3(ew_grid(n)), dim3(256), 0, stream,
                     (const float4*)x.data_ptr<float>(), (float4*)y.data_ptr<float>(),
                     x.data_ptr<float>(), y.data_ptr<float>(), n4, n);
  return y;
}

at::Tensor relu_bwd(const at::Tensor& gy, const at::Tensor& y) {
  auto gx = at::empty_like(gy);
  const long n = gy.numel();
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(relu_bwd_kernel, dim3(ew_grid(n, 256, 1)), dim3(256), 0, stream,
                     gy.data_ptr<float>(), y.data_ptr<float>(), gx.data_ptr<float>(),
                     n);
  return gx;
}

// ---------------- GELU (exact erf — matches nn.GELU default) ----------------
__global__ void gelu_fwd_kernel(const float* __restrict__ x, float* __restrict__ y,
                                long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const float v = x[i];
    y[i] = 0.5f * v * (1.f + erff(v * 0.70710678118654752f));
  }
}

__global__ void gelu_bwd_kernel(const float* __restrict__ gy,
                                const float* __restrict__ x, float* __restrict__ gx,
                                long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const float v = x[i];
    const float cdf = 0.5f * (1.f + erff(v * 0.70710678118654752f));
    const float pdf = 0.3989422804014327f * expf(-0.5f * v * v);
    gx[i] = gy[i] * (cdf + v * pdf);
  }
}

at::Tensor gelu_fwd(const at::Tensor& x) {
  auto y = at::empty_like(x);
  const long n = x.numel();
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(gelu_fwd_kernel, dim3(ew_grid(n, 256, 1)), dim3(256), 0, stream,
                     x.data_ptr<float>(), y.data_ptr<float>(), n);
  return y;
}

at::Tensor gelu_bwd(const at::Tensor& gy, const at::Tensor& x) {
  auto gx = at::empty_like(gy);
  const long n = gy.numel();
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(gelu_bwd_kernel, dim3(ew_grid(n, 256, 1)), dim3(256), 0, stream,
                     gy.data_ptr<float>(), x.data_ptr<float>(), gx.data_ptr<float>(),
                     n);
  return gx;
}

// ---------------- tanh ----------------
__global__ void tanh_fwd_kernel(const float* __restrict__ x, float* __restrict__ y,
                                long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    y[i] = tanhf(x[i]);
}

__global__ void tanh_bwd_kernel(const float* __restrict__ gy,
                                const float* __restrict__ y, float* __restrict__ gx,
                                long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    gx[i] = gy[i] * (1.f - y[i] * y[i]);
}

at::Tensor tanh_fwd(const at::Tensor& x) {
  auto y = at::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(tanh_fwd_kernel, dim3(ew_grid(x.numel(), 256, 1)), dim3(256), 0,
                     stream, x.data_ptr<float>(), y.data_ptr<float>(), x.numel());
  return y;
}

at::Tensor tanh_bwd(const at::Tensor& gy, const at::Tensor& y) {
  auto gx = at::empty_like(gy);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(tanh_bwd_kernel, dim3(ew_grid(gy.numel(), 256, 1)), dim3(256), 0,
                     stream, gy.data_ptr<float>(), y.data_ptr<float>(),
                     gx.data_ptr<float>(), gy.numel());
  return gx;
}

// ---------------- dropout (counter-based philox-style mask) ----------------
__global__ void dropout_fwd_kernel(const float* __restrict__ x, float* __restrict__ y,
                                   uint8_t* __restrict__ mask, long n, float p,
                                   float scale, uint64_t seed, uint64_t offset) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const bool keep = slk_uniform(seed, offset, (uint64_t)i) >= p;
    mask[i] = keep;
    y[i] = keep ? x[i] * scale : 0.f;
  }
}

// variant reading the offset from device memory: hipGraph-replayable (a
// captured host constant would freeze the mask; a device counter bumped by a
// captured add keeps masks advancing across replays)
__global__ void dropout_fwd_dev_kernel(const float* __restrict__ x,
                                       float* __restrict__ y,
                                       uint8_t* __restrict__ mask, long n, float p,
                                       float scale, uint64_t seed,
                                       const int64_t* __restrict__ offset_ptr) {
  const uint64_t offset = (uint64_t)(*offset_ptr);
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const bool keep = slk_uniform(seed, offset, (uint64_t)i) >= p;
    mask[i] = keep;
    y[i] = keep ? x[i] * scale : 0.f;
  }
}

__global__ void dropout_bwd_kernel(const float* __restrict__ gy,
                                   const uint8_t* __restrict__ mask,
                                   float* __restrict__ gx, long n, float scale) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    gx[i] = mask[i] ? gy[i] * scale : 0.f;
}

std::vector<at::Tensor> dropout_fwd(const at::Tensor& x, double p, int64_t seed,
                                    int64_t offset) {
  auto y = at::empty_like(x);
  auto mask = at::empty(x.sizes(), x.options().dtype(at::kByte));
  const long n = x.numel();
  const float scale = 1.0f / (1.0f - (float)p);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(dropout_fwd_kernel, dim3(ew_grid(n, 256, 1)), dim3(256), 0,
                     stream, x.data_ptr<float>(), y.data_ptr<float>(),
                     mask.data_ptr<uint8_t>(), n, (float)p, scale, (uint64_t)seed,
                     (uint64_t)offset);
  return {y, mask};
}

std::vector<at::Tensor> dropout_fwd_dev(const at::Tensor& x, double p, int64_t seed,
                                        const at::Tensor& offset) {
  auto y = at::empty_like(x);
  auto mask = at::empty(x.sizes(), x.options().dtype(at::kByte));
  const long n = x.numel();
  const float scale = 1.0f / (1.0f - (float)p);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(dropout_fwd_dev_kernel, dim3(ew_grid(n, 256, 1)), dim3(256), 0,
                     stream, x.data_ptr<float>(), y.data_ptr<float>(),
                     mask.data_ptr<uint8_t>(), n, (float)p, scale, (uint64_t)seed,
                     offset.data_ptr<int64_t>());
  return {y, mask};
}

at::Tensor dropout_bwd(const at::Tensor& gy, const at::Tensor& mask, double p) {
  auto gx = at::empty_like(gy);
  const float scale = 1.0f / (1.0f - (float)p);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(dropout_bwd_kernel, dim3(ew_grid(gy.numel(), 256, 1)), dim3(256),
                     0, stream, gy.data_ptr<float>(), mask.data_ptr<uint8_t>(),
                     gx.data_ptr<float>(), gy.numel(), scale);
  return gx;
}

// ---------------- MaxPool2d 2x2 s2 with argmax stash ----------------
__global__ void maxpool_fwd_kernel(const float* __restrict__ x, float* __restrict__ y,
                                   uint8_t* __restrict__ idx, int BC, int H, int W,
                                   int OH, int OW, FastDiv d_ohow, FastDiv d_ow) {
  const long total = (long)BC * OH * OW;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    // FastDiv: plain runtime / and % were ~25-VALU divisions per element
    const unsigned bc = d_ohow.div((unsigned)i);
    const unsigned rem = d_ohow.mod((unsigned)i, bc);
    const unsigned oh = d_ow.div(rem);
    const unsigned ow = d_ow.mod(rem, oh);
    const float* xp = x + ((long)bc * H + oh * 2) * W + ow * 2;
    float best = xp[0];
    int bi = 0;
    if (ow * 2 + 1 < W && xp[1] > best) { best = xp[1]; bi = 1; }
    if (oh * 2 + 1 < H) {
      if (xp[W] > best) { best = xp[W]; bi = 2; }
      if (ow * 2 + 1 < W && xp[W + 1] > best) { best = xp[W + 1]; bi = 3; }
    }
    y[i] = best;
    idx[i] = (uint8_t)bi;
  }
}

__global__ void maxpool_bwd_kernel(const float* __restrict__ gy,
                                   const uint8_t* __restrict__ idx,
                                   float* __restrict__ gx, int BC, int H, int W,
                                   int OH, int OW, FastDiv d_hw, FastDiv d_w) {
  const long total = (long)BC * H * W;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const unsigned bc = d_hw.div((unsigned)i);
    const unsigned rem = d_hw.mod((unsigned)i, bc);
    const unsigned h = d_w.div(rem);
    const unsigned w = d_w.mod(rem, h);
    const int oh = h >> 1, ow = w >> 1;
    float g = 0.f;
    if (oh < OH && ow < OW) {
      const long o = ((long)bc * OH + oh) * OW + ow;
      const int pos = ((h & 1) << 1) | (w & 1);
      if (idx[o] == (uint8_t)pos) g = gy[o];
    }
    gx[i] = g;
  }
}

std::vector<at::Tensor> maxpool2x2_fwd(const at::Tensor& x) {
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int OH = H / 2, OW = W / 2;
  auto y = at::empty({B, C, OH, OW}, x.options());
  auto idx = at::empty({B, C, OH, OW}, x.options().dtype(at::kByte));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const long total = (long)B * C * OH * OW;
  FastDiv d_ohow, d_ow;
  d_ohow.init(OH * OW);
  d_ow.init(OW);
  TORCH_CHECK(total <= (1l << 24), "maxpool_fwd: FastDiv range");
  hipLaunchKernelGGL(maxpool_fwd_kernel, dim3(ew_grid(total, 256, 1)), dim3(256), 0,
                     stream, x.data_ptr<float>(), y.data_ptr<float>(),
                     idx.data_ptr<uint8_t>(), B * C, H, W, OH, OW, d_ohow, d_ow);
  return {y, idx};
}

at::Tensor maxpool2x2_bwd(const at::Tensor& gy, const at::Tensor& idx, int H, int W) {
  const int B = gy.size(0), C = gy.size(1), OH = gy.size(2), OW = gy.size(3);
  auto gx = at::empty({B, C, H, W}, gy.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const long total = (long)B * C * H * W;
  FastDiv d_hw, d_w;
  d_hw.init(H * W);
  d_w.init(W);
  TORCH_CHECK(total <= (1l << 24), "maxpool_bwd: FastDiv range");
  hipLaunchKernelGGL(maxpool_bwd_kernel, dim3(ew_grid(total, 256, 1)), dim3(256), 0,
                     stream, gy.data_ptr<float>(), idx.data_ptr<uint8_t>(),
                     gx.data_ptr<float>(), B * C, H, W, OH, OW, d_hw, d_w);
  return gx;
}

// ---------------- embedding ----------------
__global__ void embedding_fwd_kernel(const int64_t* __restrict__ ids,
                                     const float* __restrict__ w,
                                     float* __restrict__ y, long n_ids, int D) {
  const long total = n_ids * D;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const long row = i / D;
    const int d = (int)(i - row * D);
    y[i] = w[(long)ids[row] * D + d];
  }
}

__global__ void embedding_bwd_kernel(const int64_t* __restrict__ ids,
                                     const float* __restrict__ gy,
                                     float* __restrict__ gw, long n_ids, int D,
                                     int padding_idx) {
  const long total = n_ids * D;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const long row = i / D;
    const int d = (int)(i - row * D);
    const int64_t id = ids[row];
    if ((int)id == padding_idx) continue;
    atomicAdd(gw + id * D + d, gy[i]);
  }
}

at::Tensor embedding_fwd(const at::Tensor& ids, const at::Tensor& w) {
  auto idsc = ids.contiguous();
  const long n_ids = ids.numel();
  const int D = w.size(1);
  auto out_sizes = ids.sizes().vec();
  out_sizes.push_back(D);
  auto y = at::empty(out_sizes, w.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(embedding_fwd_kernel, dim3(ew_grid(n_ids * D, 256, 1)),
                     dim3(256), 0, stream, idsc.data_ptr<int64_t>(),
                     w.data_ptr<float>(), y.data_ptr<float>(), n_ids, D);
  return y;
}

at::Tensor embedding_bwd(const at::Tensor& ids, const at::Tensor& gy,
                         int64_t num_embeddings, int64_t padding_idx) {
  auto idsc = ids.contiguous();
  const long n_ids = ids.numel();
  const int D = gy.size(-1);
  auto gw = at::zeros({num_embeddings, D}, gy.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(embedding_bwd_kernel, dim3(ew_grid(n_ids * D, 256, 1)),
                     dim3(256), 0, stream, idsc.data_ptr<int64_t>(),
                     gy.data_ptr<float>(), gw.data_ptr<float>(), n_ids, D,
                     (int)padding_idx);
  return gw;
}

}  // namespace slk
