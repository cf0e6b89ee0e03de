// Strided/batched fp32 MFMA GEMM + Linear forward + column-sum.
// Serves Linear fwd/bwd (reference nn.Linear sites: src/model/VGG16_CIFAR10.py:107-117),
// attention QK^T / PV batched matmuls (src/model/BERT_AGNEWS.py:66-74), and
// LoRA adapters.  Transposes are handled by stride swaps — no data movement.
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


struct StridedGather {
  const float* A;
  const float* B;
  long sAb, sAm, sAk;
  long sBb, sBk, sBn;

  // Contexts hold 32-bit OFFSETS against the functor's uniform (SGPR) base
  // pointers rather than per-context 64-bit pointers: the big-tile kernel
  // keeps 8 A-contexts live and the pointer form cost 16 VGPR of its
  // register budget (the 128x128 kernel sat at 3 waves/SIMD; see
  // tile_gemm.h).  Host guards numel < 2^31 (matmul_f32).
  struct KCtx {};
  struct ACtx { int off; bool valid; };
  struct BCtx { int off; bool valid; };

  __device__ KCtx prepK(int) const { return {}; }
  __device__ ACtx prepA(int b, int m, bool valid, int) const {
    return {(int)((long)b * sAb + (long)m * sAm), valid};
  }
  __device__ float loadA(const ACtx& c, const KCtx&, int k, bool kv) const {
    const float v = A[c.off + (long)k * sAk];
    return (c.valid & kv) ? v : 0.f;
  }
  __device__ BCtx prepB(int b, int n, bool valid, int) const {
    return {(int)((long)b * sBb + (long)n * sBn), valid};
  }
  __device__ float loadB(const BCtx& c, const KCtx&, int k, bool kv) const {
    const float v = B[c.off + (long)k * sBk];
    return (c.valid & kv) ? v : 0.f;
  }
};

struct StridedStore {
  float* C;
  long sCb;
  int N;
  const float* bias;  // per-column bias (nullable)
  bool accumulate;    // atomic accumulate (split-K or beta=1)
  __device__ void store(int b, int m, int n, float v, int ks) const {
    float* p = C + (long)b * sCb + (long)m * N + n;
    if (bias != nullptr && ks == 0) v += bias[n];
    if (accumulate) {
      atomicAdd(p, v);
    } else {
      *p = v;
    }
  }
};

static void launch_strided(const at::Tensor& a2, const at::Tensor& b2, at::Tensor& c,
                           bool ta, bool tb, const c10::optional<at::Tensor>& bias,
                           bool accumulate, int split_k) {
  const bool batched = a2.dim() == 3;
  const int nb = batched ? a2.size(0) : 1;
  const int M = ta ? a2.size(-1) : a2.size(-2);
  const int K = ta ? a2.size(-2) : a2.size(-1);
  const int N = tb ? b2.size(-2) : b2.size(-1);

  StridedGather g;
  g.A = a2.data_ptr<float>();
  g.B = b2.data_ptr<float>();
  g.sAb = batched ? a2.stride(0) : 0;
  g.sAm = ta ? a2.stride(-1) : a2.stride(-2);
  g.sAk = ta ? a2.stride(-2) : a2.stride(-1);
  g.sBb = (b2.dim() == 3) ? b2.stride(0) : 0;
  g.sBk = tb ? b2.stride(-1) : b2.stride(-2);
  g.sBn = tb ? b2.stride(-2) : b2.stride(-1);

  StridedStore st;
  st.C = c.data_ptr<float>();
  st.sCb = batched ? (long)M * N : 0;
  st.N = N;
  st.bias = bias.has_value() ? bias->data_ptr<float>() : nullptr;
  st.accumulate = accumulate || split_k > 1;

  auto stream = c10::hip::getCurrentHIPStream().stream();
  slk_launch_gemm(g, st, M, N, K, nb, split_k, stream, slk_use_big(M, N));
}

at::Tensor matmul_f32(const at::Tensor& a, const at::Tensor& b, bool ta, bool tb,
                      c10::optional<at::Tensor> out, bool accumulate) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda(), "matmul_f32: GPU tensors required");
  TORCH_CHECK(a.scalar_type() == at::kFloat && b.scalar_type() == at::kFloat,
              "matmul_f32 is fp32-only");
  TORCH_CHECK(a.dim() == b.dim() && (a.dim() == 2 || a.dim() == 3),
              "matmul_f32: 2-D or 3-D tensors, same rank");
  TORCH_CHECK(a.numel() < (1ll << 31) && b.numel() < (1ll << 31),
              "matmul_f32: int32 gather offsets (tensor too large)");
  const int M = ta ? a.size(-1) : a.size(-2);
  const int Ka = ta ? a.size(-2) : a.size(-1);
  const int Kb = tb ? b.size(-1) : b.size(-2);
  const int N = tb ? b.size(-2) : b.size(-1);
  TORCH_CHECK(Ka == Kb, "matmul_f32: inner dims mismatch ", Ka, " vs ", Kb);
  if (a.dim() == 3) TORCH_CHECK(a.size(0) == b.size(0), "batch mismatch");

  at::Tensor c;
  if (out.has_value()) {
    c = *out;
  } else {
    c = a.dim() == 3 ? at::empty({a.size(0), M, N}, a.options())
                     : at::empty({M, N}, a.options());
  }
  int split_k = 1;
  if (!accumulate && !out.has_value()) {
    const int bm = slk_use_big(M, N) ? SLK_BM2 : SLK_BM;
    const int bn = slk_use_big(M, N) ? SLK_BN2 : SLK_BN;
    split_k = slk_pick_split_k(M, N, Ka, a.dim() == 3 ? a.size(0) : 1, bm, bn);
    if (split_k > 1) {
      slk_zero_async(c.data_ptr<float>(), c.numel(),
                     c10::hip::getCurrentHIPStream().stream());
    }
  }
  launch_strided(a, b, c, ta, tb, c10::nullopt, accumulate, split_k);
  return c;
}

at::Tensor linear_fwd(const at::Tensor& x, const at::Tensor& w,
                      c10::optional<at::Tensor> bias) {
  // y[M,N] = x[M,K] @ w[N,K]^T + b; split-K when the tile grid underfills the
  // chip (batch-32 rows = 1 M-tile: the 4096->10 classifier is ONE tile).
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kFloat, "linear_fwd: fp32 GPU");
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == at::kFloat,
              "linear_fwd: weight must be a fp32 GPU tensor (got device ",
              w.device(), ")");
  TORCH_CHECK(!bias.has_value() || bias->is_cuda(), "linear_fwd: bias device");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1));
  const int M = x.size(0), N = w.size(0), K = x.size(1);
  const bool big = slk_use_big(M, N);
  int split_k = slk_pick_split_k(M, N, K, 1, big ? SLK_BM2 : SLK_BM,
                                 big ? SLK_BN2 : SLK_BN);
  auto y = split_k > 1 ? zeroed({M, N}, x.options())
                       : at::empty({M, N}, x.options());
  launch_strided(x, w, y, /*ta=*/false, /*tb=*/true, bias, /*acc=*/false, split_k);
  return y;
}

// column sum of a [M,N] matrix -> [N] (Linear bias gradient)
__global__ void colsum_kernel(const float* __restrict__ x, float* __restrict__ out,
                              int M, int N) {
  int n = blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= N) return;
  float acc = 0.f;
  for (int m = 0; m < M; ++m) acc += x[(long)m * N + n];
  out[n] = acc;
}

at::Tensor colsum_f32(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.scalar_type() == at::kFloat);
  auto xc = x.contiguous();
  auto out = at::empty({x.size(1)}, x.options());
  int N = x.size(1);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(colsum_kernel, dim3(ceil_div(N, 256)), dim3(256), 0, stream,
                     xc.data_ptr<float>(), out.data_ptr<float>(), x.size(0), N);
  return out;
}

}  // namespace slk
