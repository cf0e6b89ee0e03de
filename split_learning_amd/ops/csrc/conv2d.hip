// Implicit-GEMM NCHW fp32 convolution on the MFMA tile framework.
//
// Forward:     y[Co, (B,OH,OW)] = W[Co, (Ci,KH,KW)] @ im2col(x_pad)
// Bwd-data:    gx = valid-conv(pad(gy), flipT(W))      (stride 1)
//              gx[Ci, (B,H,W)] = W^T-gather @ col(gy)  (stride > 1 fallback)
// Bwd-weight:  gw[Co, (Ci,KH,KW)] = gy-gather @ im2col(x_pad)^T
//
// Round-2 redesign (PMC round 1: conv GEMMs = 55% of the step, MFMA only
// ~29% busy, waves issue-bound on the per-element gather math — VERDICT #1):
// boundary arithmetic is ELIMINATED from the hot loops by padding inputs in
// memory once per call (pad_nchw kernel) so every gather is a valid-conv
// (pad 0) whose addresses decompose into
//     [per-thread hoisted pointer] + [wave-uniform SALU offset]
// with NO per-element bounds compare/clamp/select chain:
// * fwd loadB:   px(n) + (ci*Hp + kh)*Wp + kw      — SALU offset, plain load;
// * bwd-data:    gx = fwd-conv of pad(gy, KH-1-p) with flip-transposed
//   weights (w^T flip kernel, ~us on <10 MB) -> same bounds-free gather;
// * bwd-weight:  x_pad gather is all-SALU; the gy (A-side) per-lane k
//   decomposition collapses via the prepK window hoist: when OH*OW % 16 == 0
//   a BK=16 window stays inside one (b, rem0) row, so b/rem decompose ONCE
//   per tile on the scalar unit (KCtx) and the per-lane address is
//   row_m + dk + kctx.a_off.
// Geometry (KH, KW, stride) stays a TEMPLATE specialisation; runtime dims
// divide through FastDiv magics (common.h); every conv GEMM split-Ks when its
// tile grid underfills the chip (fp32 atomic accumulation, first-split bias).
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

// ---------------- forward (pre-padded input: pad == 0 in-kernel) ----------
template <int CKH, int CKW, int CS>
struct ConvFwdGather {
  using G = Geo<CKH, CKW, CS, 0>;
  const float* w;  // [Co, Ci, KH, KW] (or [Ci, Co, KHKW] flipT for bwd-data)
  const float* x;  // [B, Ci, H, W] pre-padded: every tap address is in-bounds
  ConvGeom geo;

  struct KCtx {};
  __device__ KCtx prepK(int) const { return {}; }

  struct ACtx { const float* row; bool valid; };
  struct BCtx { const float* px; };

  __device__ ACtx prepA(int, int m, bool valid, int) const {
    return {w + (long)m * (geo.Ci * G::kh_kw(geo)), valid};
  }
  __device__ float loadA(const ACtx& c, const KCtx&, int k, bool kv) const {
    return sel0(c.row[k], c.valid & kv);
  }
  __device__ BCtx prepB(int, int n, bool, int) const {
    const unsigned b = geo.d_ohow.div(n);
    const unsigned rem = geo.d_ohow.mod(n, b);
    const unsigned oh = geo.d_ow.div(rem);
    const unsigned ow = geo.d_ow.mod(rem, oh);
    return {x + (long)b * geo.Ci * geo.H * geo.W
              + (long)(oh * G::stride(geo)) * geo.W + ow * G::stride(geo)};
  }
  __device__ float loadB(const BCtx& c, const KCtx&, int k, bool kv) const {
    // k is wave-uniform: the whole decomposition + offset is SALU; the load
    // is hoisted-pointer + scalar offset with NO bounds math (valid-conv)
    int ci, kh, kw;
    G::dk(geo, k, ci, kh, kw);
    const int off = (ci * geo.H + kh) * geo.W + kw;
    return sel0(c.px[off], kv);
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

// ---------------- backward data (stride > 1 fallback only) ----------------
// stride-1 bwd-data runs as a forward valid-conv on pad(gy) with flipT(w);
// this gather keeps the stride-aware scatter logic for the s=2/s=4 cases
// (MobileNet 3x3 s2, ViT 4x4 s4), operating on the UNPADDED gy.
template <int CKH, int CKW, int CS, int CP>
struct ConvBwdDataGather {
  using G = Geo<CKH, CKW, CS, CP>;
  const float* w;   // [Co, Ci, KH, KW]
  const float* gy;  // [B, Co, OH, OW]
  ConvGeom geo;

  struct KCtx {};
  __device__ KCtx prepK(int) const { return {}; }

  struct ACtx { int m; bool valid; };
  struct BCtx { const float* base; int ihp, iwp; bool valid; };

  __device__ ACtx prepA(int, int m, bool valid, int) const { return {m, valid}; }
  __device__ float loadA(const ACtx& c, const KCtx&, int k, bool kv) const {
    int co, kh, kw;
    G::dk(geo, k, co, kh, kw);
    const int khkw = G::kh_kw(geo);
    const float v = w[((long)co * geo.Ci + c.m) * khkw + kh * (G::fixed ? CKW : geo.KW) + kw];
    return sel0(v, c.valid & kv);
  }
  __device__ BCtx prepB(int, int n, bool valid, int) const {
    const unsigned b = geo.d_hw.div(n);
    const unsigned rem = geo.d_hw.mod(n, b);
    const unsigned ih = geo.d_w.div(rem);
    const unsigned iw = geo.d_w.mod(rem, ih);
    return {gy + (long)b * geo.Co * geo.OH * geo.OW,
            (int)ih + G::pad(geo), (int)iw + G::pad(geo), valid};
  }
  __device__ float loadB(const BCtx& c, const KCtx&, int k, bool kv) const {
    int co, kh, kw;
    G::dk(geo, k, co, kh, kw);
    const int s = G::stride(geo);
    const int oh_num = c.ihp - kh;
    const int ow_num = c.iwp - kw;
    int oh, ow;
    bool ok;
    if (s == 2) {
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

// ---------------- backward weight (pre-padded input) -----------------------
template <int CKH, int CKW, int CS>
struct ConvBwdWeightGather {
  using G = Geo<CKH, CKW, CS, 0>;
  const float* gy;  // [B, Co, OH, OW]
  const float* x;   // [B, Ci, H, W] pre-padded
  ConvGeom geo;
  bool fast;        // OH*OW % 16 == 0: a BK window stays inside one (b, row)

  // window hoist: k = (b, oh, ow) decomposes ONCE per tile on the SALU when
  // fast (k0 % 16 == 0 and OHOW % 16 == 0 keep b constant over the window)
  struct KCtx { long a_off; };
  __device__ KCtx prepK(int k0) const {
    if (!fast) return {0};
    const unsigned b0 = geo.d_ohow.div(k0);
    const unsigned rem0 = geo.d_ohow.mod(k0, b0);
    return {(long)b0 * geo.Co * (geo.OH * geo.OW) + rem0};
  }

  struct ACtx { const float* rowdk; int m; bool valid; };
  __device__ ACtx prepA(int, int m, bool valid, int dk) const {
    return {gy + (long)m * (geo.OH * geo.OW) + dk, m, valid};
  }
  __device__ float loadA(const ACtx& c, const KCtx& kc, int k, bool kv) const {
    float v;
    if (fast) {
      v = c.rowdk[kc.a_off];     // per-lane hoisted row + per-tile scalar
    } else {
      const unsigned b = geo.d_ohow.div(k);
      const unsigned rem = geo.d_ohow.mod(k, b);
      v = gy[((long)b * geo.Co + c.m) * (geo.OH * geo.OW) + rem];
    }
    return sel0(v, c.valid & kv);
  }

  struct BCtx { long off; };     // (ci*Hp + kh)*Wp + kw, hoisted per thread
  __device__ BCtx prepB(int, int n, bool, int) const {
    int ci, kh, kw;
    G::dk(geo, n, ci, kh, kw);
    return {((long)ci * geo.H + kh) * geo.W + kw};
  }
  __device__ float loadB(const BCtx& c, const KCtx&, int k, bool kv) const {
    // all-SALU scalar offset (k wave-uniform), bounds-free on x_pad
    const unsigned b = geo.d_ohow.div(k);
    const unsigned rem = geo.d_ohow.mod(k, b);
    const unsigned oh = geo.d_ow.div(rem);
    const unsigned ow = geo.d_ow.mod(rem, oh);
    const long sc = (long)b * geo.Ci * geo.H * geo.W
                  + (long)(oh * G::stride(geo)) * geo.W + ow * G::stride(geo);
    return sel0(x[c.off + sc], kv);
  }
};

struct AtomicStore {
  float* c;  // [M, N] contiguous, pre-zeroed
  int N;
  __device__ void store(int, int m, int n, float v, bool) const {
    atomicAdd(c + (long)m * N + n, v);
  }
};

// ---------------- padding / weight-flip prep kernels -----------------------

// x [BC, H, W] -> out [BC, H+2p, W+2p] zero-filled border, one coalesced
// pass: one blockIdx.y per (b, c) plane, blockIdx.x strides the padded plane
__global__ void pad_nchw_kernel(const float* __restrict__ x,
                                float* __restrict__ out, int H, int W, int p,
                                FastDiv d_wp) {
  const int Hp = H + 2 * p, Wp = W + 2 * p;
  const int plane = Hp * Wp;
  const float* __restrict__ xin = x + (long)blockIdx.y * H * W;
  float* __restrict__ po = out + (long)blockIdx.y * plane;
  const int stride = gridDim.x * blockDim.x;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < plane; i += stride) {
    const int hp = d_wp.div((unsigned)i);
    const int wp = d_wp.mod((unsigned)i, (unsigned)hp);
    const int ih = hp - p, iw = wp - p;
    const bool in = (unsigned)ih < (unsigned)H && (unsigned)iw < (unsigned)W;
    po[i] = in ? xin[ih * W + iw] : 0.f;
  }
}

// w [Co, Ci, KHKW] -> wt [Ci, Co, KHKW] with the tap index reversed
// (kh, kw) -> (KH-1-kh, KW-1-kw) == r -> KHKW-1-r; output-coalesced
__global__ void flipT_w_kernel(const float* __restrict__ w,
                               float* __restrict__ wt, int Co, int Ci, int KK) {
  const long total = (long)Co * Ci * KK;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int r = (int)(i % KK);
    const long t = i / KK;
    const int co = (int)(t % Co);
    const long ci = t / Co;
    wt[i] = w[((long)co * Ci + ci) * KK + (KK - 1 - r)];
  }
}

at::Tensor pad_nchw(const at::Tensor& x, int p) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.scalar_type() == at::kFloat);
  if (p == 0) return x.contiguous();
  auto xc = x.contiguous();
  const long BC = (long)x.size(0) * x.size(1);
  TORCH_CHECK(BC < 65536, "pad_nchw: B*C grid limit");
  const int H = x.size(2), W = x.size(3);
  auto out = at::empty({x.size(0), x.size(1), H + 2 * p, W + 2 * p}, x.options());
  FastDiv d_wp;
  d_wp.init(W + 2 * p);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int plane = (H + 2 * p) * (W + 2 * p);
  const int gx = std::min(ceil_div(plane, 256), 16);
  hipLaunchKernelGGL(pad_nchw_kernel, dim3(gx, (uint32_t)BC), dim3(256), 0,
                     stream, xc.data_ptr<float>(), out.data_ptr<float>(), H, W,
                     p, d_wp);
  return out;
}

static at::Tensor flipT_w(const at::Tensor& w) {
  auto wc = w.contiguous();
  const int Co = w.size(0), Ci = w.size(1), KK = w.size(2) * w.size(3);
  auto wt = at::empty({Ci, Co, w.size(2), w.size(3)}, w.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const long total = (long)Co * Ci * KK;
  const int grid = (int)std::min<long>((total + 255) / 256, 4096);
  hipLaunchKernelGGL(flipT_w_kernel, dim3(grid), dim3(256), 0, stream,
                     wc.data_ptr<float>(), wt.data_ptr<float>(), Co, Ci, KK);
  return wt;
}

// ---------------- host wrappers ----------------

// geometry over the PADDED input (pad == 0 in-kernel)
static ConvGeom make_geom_padded(int B, int Ci, int Hp, int Wp, int Co, int KH,
                                 int KW, int stride) {
  ConvGeom g;
  g.B = B; g.Ci = Ci; g.H = Hp; g.W = Wp; g.Co = Co; g.KH = KH; g.KW = KW;
  g.stride = stride; g.pad = 0;
  g.OH = (Hp - KH) / stride + 1;
  g.OW = (Wp - KW) / stride + 1;
  g.d_ohow.init(g.OH * g.OW);
  g.d_ow.init(g.OW);
  g.d_hw.init(g.H * g.W);
  g.d_w.init(g.W);
  g.d_khkw.init(g.KH * g.KW);
  g.d_kw.init(g.KW);
  return g;
}

// geometry over the ORIGINAL input (bwd-data stride>1 fallback only)
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

// dispatch over the model zoo's conv geometries (padded: stride only)
template <typename F>
static void dispatch_geom(const ConvGeom& g, F&& f) {
  if (g.KH == 3 && g.KW == 3 && g.stride == 1) {
    f(std::integral_constant<int, 0>{});  // 3x3 s1
  } else if (g.KH == 3 && g.KW == 3 && g.stride == 2) {
    f(std::integral_constant<int, 1>{});  // 3x3 s2
  } else if (g.KH == 1 && g.KW == 1 && g.stride == 1) {
    f(std::integral_constant<int, 2>{});  // 1x1
  } else if (g.KH == 4 && g.KW == 4 && g.stride == 4) {
    f(std::integral_constant<int, 3>{});  // ViT patch embed
  } else {
    f(std::integral_constant<int, 4>{});  // generic runtime geometry
  }
}

template <template <int, int, int> class Gather, int CASE>
struct PickGather;
template <template <int, int, int> class Gather>
struct PickGather<Gather, 0> { using type = Gather<3, 3, 1>; };
template <template <int, int, int> class Gather>
struct PickGather<Gather, 1> { using type = Gather<3, 3, 2>; };
template <template <int, int, int> class Gather>
struct PickGather<Gather, 2> { using type = Gather<1, 1, 1>; };
template <template <int, int, int> class Gather>
struct PickGather<Gather, 3> { using type = Gather<4, 4, 4>; };
template <template <int, int, int> class Gather>
struct PickGather<Gather, 4> { using type = Gather<0, 0, 0>; };

// x_is_padded: caller already ran pad_nchw (functional.py saves x_pad from
// the forward so the backward-weight pass reuses it without re-padding)
at::Tensor conv2d_fwd(const at::Tensor& x, const at::Tensor& w,
                      c10::optional<at::Tensor> bias, int stride, int pad,
                      bool x_is_padded) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.scalar_type() == at::kFloat);
  TORCH_CHECK(w.is_cuda() && w.dim() == 4 && w.size(1) == x.size(1));
  auto xp = x_is_padded ? x.contiguous() : pad_nchw(x, pad);
  auto wc = w.contiguous();
  ConvGeom geo = make_geom_padded(xp.size(0), xp.size(1), xp.size(2), xp.size(3),
                                  w.size(0), w.size(2), w.size(3), stride);
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
    GT g{wc.data_ptr<float>(), xp.data_ptr<float>(), geo};
    slk_launch_gemm(g, st, M, N, K, 1, split_k, stream);
  });
  return y;
}

at::Tensor conv2d_bwd_data(const at::Tensor& gy, const at::Tensor& w, int stride,
                           int pad, int H, int W) {
  TORCH_CHECK(gy.is_cuda() && gy.dim() == 4 && gy.scalar_type() == at::kFloat);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int KH = w.size(2), KW = w.size(3);
  const int Ci = w.size(1), Co = w.size(0);
  const int B = gy.size(0);

  if (stride == 1 && KH == KW && pad <= KH - 1) {
    // gx = valid-conv(pad(gy, K-1-p), flipT(w)): bounds-free forward gather.
    // Derivation: y[oh] sums x[oh - p + kh]  =>  gx[ih] = sum_kh gy[ih + p - kh]
    // = sum_kh' gy_pad[ih + kh'] with kh' = KH-1-kh and pad q = KH-1-p.
    auto wt = flipT_w(w);
    const int q = KH - 1 - pad;
    auto gyp = pad_nchw(gy, q);
    ConvGeom geo = make_geom_padded(B, Co, gyp.size(2), gyp.size(3), Ci, KH, KW, 1);
    TORCH_CHECK(geo.OH == H && geo.OW == W, "bwd-data geometry mismatch");
    const int M = Ci, N = B * H * W, K = Co * KH * KW;
    const int split_k = slk_pick_split_k(M, N, K, 1);
    auto gx = split_k > 1 ? zeroed({B, Ci, H, W}, gy.options())
                          : at::empty({B, Ci, H, W}, gy.options());
    ConvFwdStore st{gx.data_ptr<float>(), nullptr, Ci, H * W, geo.d_ohow,
                    split_k > 1};
    dispatch_geom(geo, [&](auto ic) {
      using GT = typename PickGather<ConvFwdGather, decltype(ic)::value>::type;
      GT g{wt.data_ptr<float>(), gyp.data_ptr<float>(), geo};
      slk_launch_gemm(g, st, M, N, K, 1, split_k, stream);
    });
    return gx;
  }

  // stride > 1 (or non-square taps): scatter-style gather on the unpadded gy
  auto gyc = gy.contiguous();
  auto wc = w.contiguous();
  ConvGeom geo = make_geom(B, Ci, H, W, Co, KH, KW, stride, pad);
  TORCH_CHECK(geo.OH == gy.size(2) && geo.OW == gy.size(3),
              "bwd-data geometry mismatch");
  const int M = geo.Ci, N = geo.B * H * W, K = geo.Co * KH * KW;
  const int split_k = slk_pick_split_k(M, N, K, 1);
  auto gx = split_k > 1 ? zeroed({B, Ci, H, W}, gy.options())
                        : at::empty({B, Ci, H, W}, gy.options());
  ConvBwdDataStore st{gx.data_ptr<float>(), geo.Ci, H * W, geo.d_hw,
                      split_k > 1};
  if (KH == 3 && KW == 3 && stride == 2 && pad == 1) {
    ConvBwdDataGather<3, 3, 2, 1> g{wc.data_ptr<float>(), gyc.data_ptr<float>(), geo};
    slk_launch_gemm(g, st, M, N, K, 1, split_k, stream);
  } else if (KH == 4 && KW == 4 && stride == 4 && pad == 0) {
    ConvBwdDataGather<4, 4, 4, 0> g{wc.data_ptr<float>(), gyc.data_ptr<float>(), geo};
    slk_launch_gemm(g, st, M, N, K, 1, split_k, stream);
  } else {
    ConvBwdDataGather<0, 0, 0, 0> g{wc.data_ptr<float>(), gyc.data_ptr<float>(), geo};
    slk_launch_gemm(g, st, M, N, K, 1, split_k, stream);
  }
  return gx;
}

at::Tensor conv2d_bwd_weight(const at::Tensor& gy, const at::Tensor& x, int KH,
                             int KW, int stride, int pad, bool x_is_padded) {
  TORCH_CHECK(gy.is_cuda() && x.is_cuda() && gy.scalar_type() == at::kFloat);
  auto gyc = gy.contiguous();
  auto xp = x_is_padded ? x.contiguous() : pad_nchw(x, pad);
  ConvGeom geo = make_geom_padded(xp.size(0), xp.size(1), xp.size(2), xp.size(3),
                                  gy.size(1), KH, KW, stride);
  TORCH_CHECK(geo.OH == gy.size(2) && geo.OW == gy.size(3),
              "bwd-weight geometry mismatch");
  const int M = geo.Co, N = geo.Ci * KH * KW, K = geo.B * geo.OH * geo.OW;
  auto gw = zeroed({geo.Co, geo.Ci, KH, KW}, gy.options());

  AtomicStore st{gw.data_ptr<float>(), N};
  int split_k = slk_pick_split_k(M, N, K, 1);
  const bool fast = (geo.OH * geo.OW) % SLK_BK == 0;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dispatch_geom(geo, [&](auto ic) {
    using GT = typename PickGather<ConvBwdWeightGather, decltype(ic)::value>::type;
    GT g{gyc.data_ptr<float>(), xp.data_ptr<float>(), geo, fast};
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
