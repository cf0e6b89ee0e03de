// Fused scaled-dot-product attention forward for the split-learning model
// zoo's small-sequence regime (SURVEY.md §2.4 "Attention QKV+SDPA" row:
// BERT S=128 hd=64, KWT S=99 hd=64, ViT S=65 hd=32).
//
// One workgroup per (batch*head): K and V live entirely in LDS (S<=128,
// hd<=64 -> <=66 KB), each wave sweeps query rows, and softmax(QK^T*scale)@V
// is produced in ONE kernel — no [BH,S,S] scores round-trip through HBM on
// the forward, no separate softmax/scale launches.  The probability matrix
// IS written out once (it is needed for the backward, which reuses the
// existing GEMM/softmax-bwd kernels — at these sizes the fwd fusion is the
// hot part; see profiles/SUMMARY.md "fused attention").
//
// Compute layout per query row r (wave-parallel over rows):
//   scores: lane owns columns c=lane and c=lane+64; K read column-parallel
//           from LDS at stride hd+1 (odd stride -> conflict-free), q[r][d]
//           is wave-uniform -> scalar loads from global (L2-hot);
//   softmax: wave shfl reductions (max, sum) over the <=128 per-lane values;
//   P@V:    probs transposed through a per-wave LDS row; lane owns output
//           dim d=lane; V rows broadcast-free at stride hd+1.
#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace slk {

constexpr int ATTN_MAX_S = 128;
constexpr int ATTN_MAX_HD = 64;

__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const float* __restrict__ q, const float* __restrict__ k,
    const float* __restrict__ v, float* __restrict__ out,
    float* __restrict__ probs, int S, int hd, float scale) {
  extern __shared__ float lds[];
  float* ldsK = lds;                      // [S][hd+1]
  float* ldsV = ldsK + S * (hd + 1);      // [S][hd+1]
  float* ldsP = ldsV + S * (hd + 1);      // [4][ATTN_MAX_S] per-wave scratch

  const int bh = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const long base = (long)bh * S * hd;

  for (int i = tid; i < S * hd; i += 256) {
    const int c = i / hd;           // hd is 32 or 64 here -> strength-reduced
    const int d = i - c * hd;
    ldsK[c * (hd + 1) + d] = k[base + i];
    ldsV[c * (hd + 1) + d] = v[base + i];
  }
  __syncthreads();

  float* myP = ldsP + wid * ATTN_MAX_S;
  const int c0 = lane;
  const int c1 = lane + 64;

  for (int r = wid; r < S; r += 4) {
    const float* qr = q + base + (long)r * hd;   // lane-uniform -> s_loads
    float a0 = 0.f, a1 = 0.f;
    for (int d = 0; d < hd; ++d) {
      const float qd = qr[d];
      if (c0 < S) a0 += qd * ldsK[c0 * (hd + 1) + d];
      if (c1 < S) a1 += qd * ldsK[c1 * (hd + 1) + d];
    }
    a0 = (c0 < S) ? a0 * scale : -INFINITY;
    a1 = (c1 < S) ? a1 * scale : -INFINITY;

    float m = fmaxf(a0, a1);
    for (int off = 32; off > 0; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
    const float e0 = (c0 < S) ? expf(a0 - m) : 0.f;
    const float e1 = (c1 < S) ? expf(a1 - m) : 0.f;
    float s = e0 + e1;
    for (int off = 32; off > 0; off >>= 1) s += __shfl_xor(s, off, 64);
    const float inv = 1.f / s;

    float* pr = probs + ((long)bh * S + r) * S;
    if (c0 < S) {
      myP[c0] = e0 * inv;
      pr[c0] = e0 * inv;
    }
    if (c1 < S) {
      myP[c1] = e1 * inv;
      pr[c1] = e1 * inv;
    }
    // wave-synchronous: myP written by this wave only; ds waits are implicit
    if (lane < hd) {
      float acc = 0.f;
      for (int c = 0; c < S; ++c) acc += myP[c] * ldsV[c * (hd + 1) + lane];
      out[base + (long)r * hd + lane] = acc;
    }
  }
}

std::vector<at::Tensor> attn_fwd(const at::Tensor& q, const at::Tensor& k,
                                 const at::Tensor& v, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3, "attn_fwd: [BH,S,hd] cuda tensor");
  const int BH = q.size(0), S = q.size(1), hd = q.size(2);
  TORCH_CHECK(S <= ATTN_MAX_S && hd <= ATTN_MAX_HD && (hd & (hd - 1)) == 0,
              "attn_fwd: S<=128, hd<=64 pow2 (got S=", S, " hd=", hd, ")");
  auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
  auto out = at::empty_like(qc);
  auto probs = at::empty({BH, S, S}, q.options());
  const int lds_bytes = (2 * S * (hd + 1) + 4 * ATTN_MAX_S) * sizeof(float);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(attn_fwd_kernel, dim3(BH), dim3(256), lds_bytes, stream,
                     qc.data_ptr<float>(), kc.data_ptr<float>(),
                     vc.data_ptr<float>(), out.data_ptr<float>(),
                     probs.data_ptr<float>(), S, hd, (float)scale);
  return {out, probs};
}

}  // namespace slk
