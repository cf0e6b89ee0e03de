// Implicit-GEMM NCHW fp32 convolution on the MFMA tile framework.
//
// Forward:     y[Co, (B,OH,OW)] = W[Co, (Ci,KH,KW)] @ im2col(x)
// Bwd-data:    gx[Ci, (B,H,W)]  = W^T-gather @ col(gy)   (stride-aware)
// Bwd-weight:  gw[Co, (Ci,KH,KW)] = gy-gather @ im2col(x)^T
//
// Performance notes (profiled on MI355X, profiles/):
// * kernel geometry (KH, KW, stride, pad) is a TEMPLATE specialisation for the
//   model zoo's cases — 3x3 s1 p1, 3x3 s2 p1, 1x1, 4x4 s4 (ViT patch) — so
//   every k-decomposition division is strength-reduced; runtime dims (OW,
//   OH*OW, H*W) divide through FastDiv magics (common.h);
// * gathers follow the tile framework's prep/load contract: the n/m-side
//   decomposition is hoisted into per-thread contexts, and the per-element
//   loads are BRANCHLESS (clamped addresses + cndmask selects) — the earlier
//   bounds-branch version spent 53% of wave cycles issue-stalled behind
//   s_and_saveexec chains;
// * every conv GEMM split-Ks when its tile grid underfills the chip, with
//   fp32 atomic accumulation and first-split-gated bias.
#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "tile_gemm.h"

namespace slk {

// empty + single fill-KERNEL launch instead of at::zeros (no aten dispatch;
// and never hipMemsetAsync — see slk_zero_async in common.h).
static inline at::Tensor zeroed(at::IntArrayRef sizes, const at::TensorOptions& opt) {
  auto t = at::empty(sizes, opt);
  slk_zero_async(t.data_ptr<float>(), t.numel(),
                 c10::hip::getCurrentHIPStream().stream());
  return t;
}


struct ConvGeom {
  int B, Ci, H, W, Co, KH, KW, OH, OW, stride, pad;
  FastDiv d_ohow, d_ow, d_hw, d_w, d_khkw, d_kw;  // runtime-dim magics
};

// compile-time geometry: CKH=0 means runtime (generic fallback)
template <int CKH, int CKW, int CS, int CP>
struct Geo {
  static constexpr bool fixed = CKH > 0;
  __device__ static int kh_kw(const ConvGeom& g) { return fixed ? CKH * CKW : g.KH * g.KW; }
  __device__ static int stride(const ConvGeom& g) { return fixed ? CS : g.stride; }
  __device__ static int pad(const ConvGeom& g) { return fixed ? CP : g.pad; }
  // k -> (c, kh, kw) with strength-reduced division when fixed
  __device__ static void dk(const ConvGeom& g, int k, int& c, int& kh, int& kw) {
    if (fixed) {
      c = k / (CKH * CKW);
      const int r = k - c * (CKH * CKW);
      kh = r / CKW;
      kw = r - kh * CKW;
    } else {
      c = g.d_khkw.div(k);
      const int r = g.d_khkw.mod(k, c);
      kh = g.d_kw.div(r);
      kw = g.d_kw.mod(r, kh);
    }
  }
};

__device__ __forceinline__ float sel0(float v, bool keep) { return keep ? v : 0.f; }

// ---------------- forward ----------------
template <int CKH, int CKW, int CS, int CP>
struct ConvFwdGather {
  using G = Geo<CKH, CKW, CS, CP>;
  const float* w;  // [Co, Ci, KH, KW]
  const float* x;  // [B, Ci, H, W]
  ConvGeom geo;

  struct ACtx { const float* row; bool valid; };
  struct BCtx { const float* base; int ihb, iwb; bool valid; };

  __device__ ACtx prepA(int, int m, bool valid) const {
    return {w + (long)m * (geo.Ci * G::kh_kw(geo)), valid};
  }
  __device__ float loadA(const ACtx& c, int k, bool kv) const {
    return sel0(c.row[k], c.valid & kv);
  }
  __device__ BCtx prepB(int, int n, bool valid) const {
    const unsigned b = geo.d_ohow.div(n);
    const unsigned rem = geo.d_ohow.mod(n, b);
    const unsigned oh = geo.d_ow.div(rem);
    const unsigned ow = geo.d_ow.mod(rem, oh);
    return {x + (long)b * geo.Ci * geo.H * geo.W,
            (int)oh * G::stride(geo) - G::pad(geo),
            (int)ow * G::stride(geo) - G::pad(geo), valid};
  }
  __device__ float loadB(const BCtx& c, int k, bool kv) const {
    int ci, kh, kw;
    G::dk(geo, k, ci, kh, kw);
    const int ih = c.ihb + kh;
    const int iw = c.iwb + kw;
    const bool in = (unsigned)ih < (unsigned)geo.H && (unsigned)iw < (unsigned)geo.W;
    const int ihc = in ? ih : 0;
    const int iwc = in ? iw : 0;
    const float v = c.base[((long)ci * geo.H + ihc) * geo.W + iwc];
    return sel0(v, c.valid & kv & in);
  }
};

struct ConvFwdStore {
  float* y;  // [B, Co, OH, OW]
  const float* bias;  // [Co] nullable
  int Co, OHOW;
  FastDiv d_ohow;
  bool accumulate;
  __device__ void store(int, int m, int n, float v, bool first_split) const {
    const unsigned b = d_ohow.div(n);
    const unsigned rem = d_ohow.mod(n, b);
    if (bias != nullptr && first_split) v += bias[m];
    float* p = y + ((long)b * Co + m) * OHOW + rem;
    if (accumulate) {
      atomicAdd(p, v);
    } else {
      *p = v;
    }
  }
};

// ---------------- backward data ----------------
template <int CKH, int CKW, int CS, int CP>
struct ConvBwdDataGather {
  using G = Geo<CKH, CKW, CS, CP>;
  const float* w;   // [Co, Ci, KH, KW]
  const float* gy;  // [B, Co, OH, OW]
  ConvGeom geo;

  struct ACtx { int m; bool valid; };
  struct BCtx { const float* base; int ihp, iwp; bool valid; };

  __device__ ACtx prepA(int, int m, bool valid) const { return {m, valid}; }
  __device__ float loadA(const ACtx& c, int k, bool kv) const {
    int co, kh, kw;
    G::dk(geo, k, co, kh, kw);
    const int khkw = G::kh_kw(geo);
    const float v = w[((long)co * geo.Ci + c.m) * khkw + kh * (G::fixed ? CKW : geo.KW) + kw];
    return sel0(v, c.valid & kv);
  }
  __device__ BCtx prepB(int, int n, bool valid) const {
    const unsigned b = geo.d_hw.div(n);
    const unsigned rem = geo.d_hw.mod(n, b);
    const unsigned ih = geo.d_w.div(rem);
    const unsigned iw = geo.d_w.mod(rem, ih);
    return {gy + (long)b * geo.Co * geo.OH * geo.OW,
            (int)ih + G::pad(geo), (int)iw + G::pad(geo), valid};
  }
  __device__ float loadB(const BCtx& c, int k, bool kv) const {
    int co, kh, kw;
    G::dk(geo, k, co, kh, kw);
    const int s = G::stride(geo);
    const int oh_num = c.ihp - kh;
    const int ow_num = c.iwp - kw;
    int oh, ow;
    bool ok;
    if (s == 1) {
      oh = oh_num;
      ow = ow_num;
      ok = true;
    } else if (s == 2) {
      ok = ((oh_num | ow_num) & 1) == 0;
      oh = oh_num >> 1;
      ow = ow_num >> 1;
    } else if (s == 4) {
      ok = ((oh_num | ow_num) & 3) == 0;
      oh = oh_num >> 2;
      ow = ow_num >> 2;
    } else {
      ok = (oh_num % s) == 0 && (ow_num % s) == 0;
      oh = oh_num / s;
      ow = ow_num / s;
    }
    ok = ok && (unsigned)oh < (unsigned)geo.OH && (unsigned)ow < (unsigned)geo.OW;
    const int ohc = ok ? oh : 0;
    const int owc = ok ? ow : 0;
    const float v = c.base[((long)co * geo.OH + ohc) * geo.OW + owc];
    return sel0(v, c.valid & kv & ok);
  }
};

struct ConvBwdDataStore {
  float* gx;  // [B, Ci, H, W]
  int Ci, HW;
  FastDiv d_hw;
  bool accumulate;
  __device__ void store(int, int m, int n, float v, bool) const {
    const unsigned b = d_hw.div(n);
    const unsigned rem = d_hw.mod(n, b);
    float* p = gx + ((long)b * Ci + m) * HW + rem;
    if (accumulate) {
      atomicAdd(p, v);
    } else {
      *p = v;
    }
  }
};

// ---------------- backward weight ----------------
template <int CKH, int CKW, int CS, int CP>
struct ConvBwdWeightGather {
  using G = Geo<CKH, CKW, CS, CP>;
  const float* gy;  // [B, Co, OH, OW]
  const float* x;   // [B, Ci, H, W]
  ConvGeom geo;

  struct ACtx { int m; bool valid; };
  struct BCtx { long ciHW; int kh, kw; bool valid; };

  __device__ ACtx prepA(int, int m, bool valid) const { return {m, valid}; }
  __device__ float loadA(const ACtx& c, int k, bool kv) const {
    // k = (b, oh, ow): runtime-size decomposition via FastDiv
    const unsigned b = geo.d_ohow.div(k);
    const unsigned rem = geo.d_ohow.mod(k, b);
    const float v = gy[((long)b * geo.Co + c.m) * (geo.OH * geo.OW) + rem];
    return sel0(v, c.valid & kv);
  }
  __device__ BCtx prepB(int, int n, bool valid) const {
    int ci, kh, kw;
    G::dk(geo, n, ci, kh, kw);
    return {(long)ci * geo.H * geo.W, kh, kw, valid};
  }
  __device__ float loadB(const BCtx& c, int k, bool kv) const {
    const unsigned b = geo.d_ohow.div(k);
    const unsigned rem = geo.d_ohow.mod(k, b);
    const unsigned oh = geo.d_ow.div(rem);
    const unsigned ow = geo.d_ow.mod(rem, oh);
    const int ih = (int)oh * G::stride(geo) - G::pad(geo) + c.kh;
    const int iw = (int)ow * G::stride(geo) - G::pad(geo) + c.kw;
    const bool in = (unsigned)ih < (unsigned)geo.H && (unsigned)iw < (unsigned)geo.W;
    const int ihc = in ? ih : 0;
    const int iwc = in ? iw : 0;
    const float v = x[(long)b * geo.Ci * geo.H * geo.W + c.ciHW
                      + (long)ihc * geo.W + iwc];
    return sel0(v, c.valid & kv & in);
  }
};

struct AtomicStore {
  float* c;  // [M, N] contiguous, pre-zeroed
  int N;
  __device__ void store(int, int m, int n, float v, bool) const {
    atomicAdd(c + (long)m * N + n, v);
  }
};

// ---------------- host wrappers ----------------

static ConvGeom make_geom(int B, int Ci, int H, int W, int Co, int KH, int KW,
                          int stride, int pad) {
  ConvGeom g;
  g.B = B; g.Ci = Ci; g.H = H; g.W = W; g.Co = Co; g.KH = KH; g.KW = KW;
  g.stride = stride; g.pad = pad;
  g.OH = (H + 2 * pad - KH) / stride + 1;
  g.OW = (W + 2 * pad - KW) / stride + 1;
  g.d_ohow.init(g.OH * g.OW);
  g.d_ow.init(g.OW);
  g.d_hw.init(g.H * g.W);
  g.d_w.init(g.W);
  g.d_khkw.init(g.KH * g.KW);
  g.d_kw.init(g.KW);
  return g;
}

// dispatch over the model zoo's conv geometries
template <typename F>
static void dispatch_geom(const ConvGeom& g, F&& f) {
  if (g.KH == 3 && g.KW == 3 && g.stride == 1 && g.pad == 1) {
    f(std::integral_constant<int, 0>{});  // 3x3 s1 p1
  } else if (g.KH == 3 && g.KW == 3 && g.stride == 2 && g.pad == 1) {
    f(std::integral_constant<int, 1>{});  // 3x3 s2 p1
  } else if (g.KH == 1 && g.KW == 1 && g.stride == 1 && g.pad == 0) {
    f(std::integral_constant<int, 2>{});  // 1x1
  } else if (g.KH == 4 && g.KW == 4 && g.stride == 4 && g.pad == 0) {
    f(std::integral_constant<int, 3>{});  // ViT patch embed
  } else {
    f(std::integral_constant<int, 4>{});  // generic runtime geometry
  }
}

template <template <int, int, int, int> class Gather, int CASE>
struct PickGather;
template <template <int, int, int, int> class Gather>
struct PickGather<Gather, 0> { using type = Gather<3, 3, 1, 1>; };
template <template <int, int, int, int> class Gather>
struct PickGather<Gather, 1> { using type = Gather<3, 3, 2, 1>; };
template <template <int, int, int, int> class Gather>
struct PickGather<Gather, 2> { using type = Gather<1, 1, 1, 0>; };
template <template <int, int, int, int> class Gather>
struct PickGather<Gather, 3> { using type = Gather<4, 4, 4, 0>; };
template <template <int, int, int, int> class Gather>
struct PickGather<Gather, 4> { using type = Gather<0, 0, 0, 0>; };

at::Tensor conv2d_fwd(const at::Tensor& x, const at::Tensor& w,
                      c10::optional<at::Tensor> bias, int stride, int pad) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.scalar_type() == at::kFloat);
  TORCH_CHECK(w.is_cuda() && w.dim() == 4 && w.size(1) == x.size(1));
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  ConvGeom geo = make_geom(x.size(0), x.size(1), x.size(2), x.size(3),
                           w.size(0), w.size(2), w.size(3), stride, pad);
  const int M = geo.Co, N = geo.B * geo.OH * geo.OW, K = geo.Ci * geo.KH * geo.KW;
  const int split_k = slk_pick_split_k(M, N, K, 1);
  auto y = split_k > 1
      ? zeroed({geo.B, geo.Co, geo.OH, geo.OW}, x.options())
      : at::empty({geo.B, geo.Co, geo.OH, geo.OW}, x.options());

  ConvFwdStore st{y.data_ptr<float>(),
                  bias.has_value() ? bias->data_ptr<float>() : nullptr,
                  geo.Co, geo.OH * geo.OW, geo.d_ohow, split_k > 1};
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dispatch_geom(geo, [&](auto ic) {
    using GT = typename PickGather<ConvFwdGather, decltype(ic)::value>::type;
    GT g{wc.data_ptr<float>(), xc.data_ptr<float>(), geo};
    slk_launch_gemm(g, st, M, N, K, 1, split_k, stream);
  });
  return y;
}

at::Tensor conv2d_bwd_data(const at::Tensor& gy, const at::Tensor& w, int stride,
                           int pad, int H, int W) {
  TORCH_CHECK(gy.is_cuda() && gy.dim() == 4 && gy.scalar_type() == at::kFloat);
  auto gyc = gy.contiguous();
  auto wc = w.contiguous();
  ConvGeom geo = make_geom(gy.size(0), w.size(1), H, W, gy.size(1), w.size(2),
                           w.size(3), stride, pad);
  const int M = geo.Ci, N = geo.B * geo.H * geo.W, K = geo.Co * geo.KH * geo.KW;
  const int split_k = slk_pick_split_k(M, N, K, 1);
  auto gx = split_k > 1 ? zeroed({geo.B, geo.Ci, geo.H, geo.W}, gy.options())
                        : at::empty({geo.B, geo.Ci, geo.H, geo.W}, gy.options());

  ConvBwdDataStore st{gx.data_ptr<float>(), geo.Ci, geo.H * geo.W, geo.d_hw,
                      split_k > 1};
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dispatch_geom(geo, [&](auto ic) {
    using GT = typename PickGather<ConvBwdDataGather, decltype(ic)::value>::type;
    GT g{wc.data_ptr<float>(), gyc.data_ptr<float>(), geo};
    slk_launch_gemm(g, st, M, N, K, 1, split_k, stream);
  });
  return gx;
}

at::Tensor conv2d_bwd_weight(const at::Tensor& gy, const at::Tensor& x, int KH,
                             int KW, int stride, int pad) {
  TORCH_CHECK(gy.is_cuda() && x.is_cuda() && gy.scalar_type() == at::kFloat);
  auto gyc = gy.contiguous();
  auto xc = x.contiguous();
  ConvGeom geo = make_geom(x.size(0), x.size(1), x.size(2), x.size(3), gy.size(1),
                           KH, KW, stride, pad);
  const int M = geo.Co, N = geo.Ci * KH * KW, K = geo.B * geo.OH * geo.OW;
  auto gw = zeroed({geo.Co, geo.Ci, KH, KW}, gy.options());

  AtomicStore st{gw.data_ptr<float>(), N};
  int split_k = slk_pick_split_k(M, N, K, 1);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dispatch_geom(geo, [&](auto ic) {
    using GT = typename PickGather<ConvBwdWeightGather, decltype(ic)::value>::type;
    GT g{gyc.data_ptr<float>(), xc.data_ptr<float>(), geo};
    slk_launch_gemm(g, st, M, N, K, 1, split_k, stream);
  });
  return gw;
}

// per-channel sum of gy over (B, OH, OW) -> conv bias gradient
// (chunk slabs + finalize reduce: no zero-init, no atomics)
__global__ void conv_bias_grad_kernel(const float* __restrict__ gy,
                                      float* __restrict__ slab, int B, int C,
                                      int HW) {
  __shared__ float scratch[16];
  const int c = blockIdx.x;
  const int total = B * HW;
  const int per = (total + gridDim.y - 1) / gridDim.y;
  const int lo = blockIdx.y * per;
  const int hi = min(total, lo + per);
  float acc = 0.f;
  for (int i = lo + threadIdx.x; i < hi; i += blockDim.x) {
    const int b = i / HW;
    const int r = i - b * HW;
    acc += gy[((long)b * C + c) * HW + r];
  }
  float total_s = slk_block_sum(acc, scratch);
  if (threadIdx.x == 0) slab[(long)blockIdx.y * C + c] = total_s;
}

__global__ void conv_bias_finalize_kernel(const float* __restrict__ slab,
                                          int chunks, float* __restrict__ gb,
                                          int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = 0.f;
  for (int k = 0; k < chunks; ++k) s += slab[(long)k * C + c];
  gb[c] = s;
}

at::Tensor conv2d_bwd_bias(const at::Tensor& gy) {
  auto gyc = gy.contiguous();
  const int B = gy.size(0), C = gy.size(1), HW = gy.size(2) * gy.size(3);
  auto gb = at::empty({C}, gy.options());
  long chunks = ((long)B * HW) / 4096;
  if (chunks < 1) chunks = 1;
  if (chunks > 16) chunks = 16;
  auto slab = at::empty({chunks, C}, gy.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(conv_bias_grad_kernel, dim3(C, (uint32_t)chunks), dim3(256),
                     0, stream, gyc.data_ptr<float>(), slab.data_ptr<float>(), B, C,
                     HW);
  hipLaunchKernelGGL(conv_bias_finalize_kernel, dim3(ceil_div(C, 256)), dim3(256),
                     0, stream, slab.data_ptr<float>(), (int)chunks,
                     gb.data_ptr<float>(), C);
  return gb;
}

}  // namespace slk
