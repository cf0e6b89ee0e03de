// Implicit-GEMM NCHW fp32 convolution on the MFMA tile framework.
//
// Forward:     y[Co, (B,OH,OW)] = W[Co, (Ci,KH,KW)] @ im2col(x)
// Bwd-data:    gx[Ci, (B,H,W)]  = W^T-gather @ col(gy)   (stride-aware)
// Bwd-weight:  gw[Co, (Ci,KH,KW)] = gy-gather @ im2col(x)^T, split-K over
//              (B,OH,OW) with atomic accumulation (small-spatial tails need it:
//              SURVEY.md §7 hard-part 1).
// Covers every conv in the model zoo: 3x3 s1/s2 p1, 1x1, and ViT's 4x4 s4
// patch embed (reference conv sites: src/model/VGG16_CIFAR10.py:10-94,
// other/Vanilla_SL/src/model/MobileNetv1_CIFAR10.py, ViT_CIFAR10.py:44).
#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "tile_gemm.h"

namespace slk {

struct ConvGeom {
  int B, Ci, H, W, Co, KH, KW, OH, OW, stride, pad;
};

// ---------------- forward ----------------
struct ConvFwdGather {
  const float* w;  // [Co, Ci, KH, KW]
  const float* x;  // [B, Ci, H, W]
  ConvGeom geo;
  __device__ float loadA(int, int m, int k) const {  // A[Co][Ci*KH*KW]
    return w[(long)m * (geo.Ci * geo.KH * geo.KW) + k];
  }
  __device__ float loadB(int, int k, int n) const {  // B[Ci*KH*KW][B*OH*OW]
    const int ohow = geo.OH * geo.OW;
    const int b = n / ohow;
    const int rem = n - b * ohow;
    const int oh = rem / geo.OW;
    const int ow = rem - oh * geo.OW;
    const int ci = k / (geo.KH * geo.KW);
    const int r = k - ci * (geo.KH * geo.KW);
    const int kh = r / geo.KW;
    const int kw = r - kh * geo.KW;
    const int ih = oh * geo.stride - geo.pad + kh;
    const int iw = ow * geo.stride - geo.pad + kw;
    if (ih < 0 || ih >= geo.H || iw < 0 || iw >= geo.W) return 0.f;
    return x[(((long)b * geo.Ci + ci) * geo.H + ih) * geo.W + iw];
  }
};

struct ConvFwdStore {
  float* y;  // [B, Co, OH, OW]
  const float* bias;  // [Co] nullable
  int Co, OHOW;
  __device__ void store(int, int m, int n, float v) const {
    const int b = n / OHOW;
    const int rem = n - b * OHOW;
    if (bias != nullptr) v += bias[m];
    y[((long)b * Co + m) * OHOW + rem] = v;
  }
};

// ---------------- backward data ----------------
struct ConvBwdDataGather {
  const float* w;   // [Co, Ci, KH, KW]
  const float* gy;  // [B, Co, OH, OW]
  ConvGeom geo;
  __device__ float loadA(int, int m, int k) const {  // A[Ci][Co*KH*KW]
    const int khkw = geo.KH * geo.KW;
    const int co = k / khkw;
    const int r = k - co * khkw;
    return w[((long)co * geo.Ci + m) * khkw + r];
  }
  __device__ float loadB(int, int k, int n) const {  // B[Co*KH*KW][B*H*W]
    const int hw = geo.H * geo.W;
    const int b = n / hw;
    const int rem = n - b * hw;
    const int ih = rem / geo.W;
    const int iw = rem - ih * geo.W;
    const int khkw = geo.KH * geo.KW;
    const int co = k / khkw;
    const int r = k - co * khkw;
    const int kh = r / geo.KW;
    const int kw = r - kh * geo.KW;
    const int oh_num = ih + geo.pad - kh;
    const int ow_num = iw + geo.pad - kw;
    if (oh_num < 0 || ow_num < 0) return 0.f;
    if (oh_num % geo.stride != 0 || ow_num % geo.stride != 0) return 0.f;
    const int oh = oh_num / geo.stride;
    const int ow = ow_num / geo.stride;
    if (oh >= geo.OH || ow >= geo.OW) return 0.f;
    return gy[(((long)b * geo.Co + co) * geo.OH + oh) * geo.OW + ow];
  }
};

struct ConvBwdDataStore {
  float* gx;  // [B, Ci, H, W]
  int Ci, HW;
  __device__ void store(int, int m, int n, float v) const {
    const int b = n / HW;
    const int rem = n - b * HW;
    gx[((long)b * Ci + m) * HW + rem] = v;
  }
};

// ---------------- backward weight ----------------
struct ConvBwdWeightGather {
  const float* gy;  // [B, Co, OH, OW]
  const float* x;   // [B, Ci, H, W]
  ConvGeom geo;
  __device__ float loadA(int, int m, int k) const {  // A[Co][B*OH*OW]
    const int ohow = geo.OH * geo.OW;
    const int b = k / ohow;
    const int rem = k - b * ohow;
    return gy[((long)b * geo.Co + m) * ohow + rem];
  }
  __device__ float loadB(int, int k, int n) const {  // B[B*OH*OW][Ci*KH*KW]
    const int ohow = geo.OH * geo.OW;
    const int b = k / ohow;
    const int rem = k - b * ohow;
    const int oh = rem / geo.OW;
    const int ow = rem - oh * geo.OW;
    const int khkw = geo.KH * geo.KW;
    const int ci = n / khkw;
    const int r = n - ci * khkw;
    const int kh = r / geo.KW;
    const int kw = r - kh * geo.KW;
    const int ih = oh * geo.stride - geo.pad + kh;
    const int iw = ow * geo.stride - geo.pad + kw;
    if (ih < 0 || ih >= geo.H || iw < 0 || iw >= geo.W) return 0.f;
    return x[(((long)b * geo.Ci + ci) * geo.H + ih) * geo.W + iw];
  }
};

struct AtomicStore {
  float* c;  // [M, N] contiguous, pre-zeroed
  int N;
  __device__ void store(int, int m, int n, float v) const {
    atomicAdd(c + (long)m * N + n, v);
  }
};

// ---------------- host wrappers ----------------

static ConvGeom make_geom(const at::Tensor& x, const at::Tensor& w, int stride, int pad) {
  ConvGeom g;
  g.B = x.size(0); g.Ci = x.size(1); g.H = x.size(2); g.W = x.size(3);
  g.Co = w.size(0); g.KH = w.size(2); g.KW = w.size(3);
  g.stride = stride; g.pad = pad;
  g.OH = (g.H + 2 * pad - g.KH) / stride + 1;
  g.OW = (g.W + 2 * pad - g.KW) / stride + 1;
  return g;
}

at::Tensor conv2d_fwd(const at::Tensor& x, const at::Tensor& w,
                      c10::optional<at::Tensor> bias, int stride, int pad) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.scalar_type() == at::kFloat);
  TORCH_CHECK(w.is_cuda() && w.dim() == 4 && w.size(1) == x.size(1));
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  ConvGeom geo = make_geom(xc, wc, stride, pad);
  auto y = at::empty({geo.B, geo.Co, geo.OH, geo.OW}, x.options());

  ConvFwdGather g{wc.data_ptr<float>(), xc.data_ptr<float>(), geo};
  ConvFwdStore st{y.data_ptr<float>(),
                  bias.has_value() ? bias->data_ptr<float>() : nullptr,
                  geo.Co, geo.OH * geo.OW};
  const int M = geo.Co, N = geo.B * geo.OH * geo.OW, K = geo.Ci * geo.KH * geo.KW;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  slk_launch_gemm(g, st, M, N, K, 1, 1, stream);
  return y;
}

at::Tensor conv2d_bwd_data(const at::Tensor& gy, const at::Tensor& w, int stride,
                           int pad, int H, int W) {
  TORCH_CHECK(gy.is_cuda() && gy.dim() == 4 && gy.scalar_type() == at::kFloat);
  auto gyc = gy.contiguous();
  auto wc = w.contiguous();
  ConvGeom geo;
  geo.B = gy.size(0); geo.Co = gy.size(1); geo.OH = gy.size(2); geo.OW = gy.size(3);
  geo.Ci = w.size(1); geo.KH = w.size(2); geo.KW = w.size(3);
  geo.H = H; geo.W = W; geo.stride = stride; geo.pad = pad;
  auto gx = at::empty({geo.B, geo.Ci, geo.H, geo.W}, gy.options());

  ConvBwdDataGather g{wc.data_ptr<float>(), gyc.data_ptr<float>(), geo};
  ConvBwdDataStore st{gx.data_ptr<float>(), geo.Ci, geo.H * geo.W};
  const int M = geo.Ci, N = geo.B * geo.H * geo.W, K = geo.Co * geo.KH * geo.KW;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  slk_launch_gemm(g, st, M, N, K, 1, 1, stream);
  return gx;
}

at::Tensor conv2d_bwd_weight(const at::Tensor& gy, const at::Tensor& x, int KH,
                             int KW, int stride, int pad) {
  TORCH_CHECK(gy.is_cuda() && x.is_cuda() && gy.scalar_type() == at::kFloat);
  auto gyc = gy.contiguous();
  auto xc = x.contiguous();
  ConvGeom geo;
  geo.B = x.size(0); geo.Ci = x.size(1); geo.H = x.size(2); geo.W = x.size(3);
  geo.Co = gy.size(1); geo.OH = gy.size(2); geo.OW = gy.size(3);
  geo.KH = KH; geo.KW = KW; geo.stride = stride; geo.pad = pad;

  const int M = geo.Co, N = geo.Ci * KH * KW, K = geo.B * geo.OH * geo.OW;
  auto gw = at::zeros({geo.Co, geo.Ci, KH, KW}, gy.options());

  ConvBwdWeightGather g{gyc.data_ptr<float>(), xc.data_ptr<float>(), geo};
  AtomicStore st{gw.data_ptr<float>(), N};
  int split_k = slk_pick_split_k(M, N, K, 1);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  slk_launch_gemm(g, st, M, N, K, 1, split_k, stream);
  return gw;
}

// per-channel sum of gy over (B, OH, OW) -> conv bias gradient
__global__ void conv_bias_grad_kernel(const float* __restrict__ gy,
                                      float* __restrict__ gb, int B, int C, int HW) {
  __shared__ float scratch[16];
  const int c = blockIdx.x;
  float acc = 0.f;
  for (int i = threadIdx.x; i < B * HW; i += blockDim.x) {
    const int b = i / HW;
    const int r = i - b * HW;
    acc += gy[((long)b * C + c) * HW + r];
  }
  float total = slk_block_sum(acc, scratch);
  if (threadIdx.x == 0) gb[c] = total;
}

at::Tensor conv2d_bwd_bias(const at::Tensor& gy) {
  auto gyc = gy.contiguous();
  const int B = gy.size(0), C = gy.size(1), HW = gy.size(2) * gy.size(3);
  auto gb = at::empty({C}, gy.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(conv_bias_grad_kernel, dim3(C), dim3(256), 0, stream,
                     gyc.data_ptr<float>(), gb.data_ptr<float>(), B, C, HW);
  return gb;
}

}  // namespace slk
