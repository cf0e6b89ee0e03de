// Fused scaled-dot-product attention forward for the split-learning model
// zoo's small-sequence regime (SURVEY.md §2.4 "Attention QKV+SDPA" row:
// BERT S=128 hd=64, KWT S=99 hd=64, ViT S=65 hd=32).
//
// One workgroup per (batch*head): K and V live entirely in LDS (S<=128,
// hd<=64 -> <=66 KB), each wave sweeps query rows, and
// dropout(softmax(QK^T*scale))@V is produced in ONE kernel — no [BH,S,S]
// scores round-trip through HBM on the forward, no separate softmax/scale/
// dropout launches.  Round 2: attention DROPOUT runs inside the kernel
// (counter-based splitmix RNG, common.h), so BERT's p=0.1 training uses the
// fused path (round-1 VERDICT missing #3).  The kernel writes the CLEAN
// probability matrix plus the dropout mask; the backward reuses the
// GEMM/softmax-bwd/dropout-bwd kernels:
//   pd = P.mask/(1-p); dV = pd^T@gO; dP = (gO@V^T).mask/(1-p);
//   dS = softmax_bwd(dP, P)*scale; dQ = dS@K; dK = dS^T@Q.
//
// Compute layout per query row r (wave-parallel over rows):
//   scores: lane owns columns c=lane and c=lane+64; K read column-parallel
//           from LDS at stride hd+1 (odd stride -> conflict-free), q[r][d]
//           is wave-uniform -> scalar loads from global (L2-hot);
//   softmax: wave shfl reductions (max, sum) over the <=128 per-lane values;
//   P@V:    (dropped) probs transposed through a per-wave LDS row; lane owns
//           output dim d=lane; V rows broadcast-free at stride hd+1.
#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace slk {

constexpr int ATTN_MAX_S = 128;
constexpr int ATTN_MAX_HD = 64;

template <bool DROPOUT>
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const float* __restrict__ q, const float* __restrict__ k,
    const float* __restrict__ v, float* __restrict__ out,
    float* __restrict__ probs, unsigned char* __restrict__ mask, int S, int hd,
    float scale, float p, float inv_keep, uint64_t seed,
    const long* __restrict__ offset_ptr) {
  // device-side philox offset (graph-safe: advances under hipGraph replay),
  // folded with the seed into one 32-bit stream id for the cheap mask hash
  const uint64_t offset = offset_ptr ? (uint64_t)offset_ptr[0] : 0;
  const uint32_t rng_stream = (uint32_t)(slk_mix64(seed ^ slk_mix64(offset)) >> 32);
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
    const long ridx = ((long)bh * S + r) * S;
    if (c0 < S) {
      const float pv = e0 * inv;
      float pd = pv;
      if (DROPOUT) {
        const bool keep = slk_uniform32(rng_stream, (uint32_t)(ridx + c0)) >= p;
        pd = keep ? pv * inv_keep : 0.f;
        mask[ridx + c0] = keep;
      }
      myP[c0] = pd;
      pr[c0] = pv;                  // CLEAN probs (softmax bwd needs them)
    }
    if (c1 < S) {
      const float pv = e1 * inv;
      float pd = pv;
      if (DROPOUT) {
        const bool keep = slk_uniform32(rng_stream, (uint32_t)(ridx + c1)) >= p;
        pd = keep ? pv * inv_keep : 0.f;
        mask[ridx + c1] = keep;
      }
      myP[c1] = pd;
      pr[c1] = pv;
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
                                 const at::Tensor& v, double scale, double p,
                                 int64_t seed, c10::optional<at::Tensor> offset) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3, "attn_fwd: [BH,S,hd] cuda tensor");
  const int BH = q.size(0), S = q.size(1), hd = q.size(2);
  TORCH_CHECK(S <= ATTN_MAX_S && hd <= ATTN_MAX_HD && (hd & (hd - 1)) == 0,
              "attn_fwd: S<=128, hd<=64 pow2 (got S=", S, " hd=", hd, ")");
  auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
  auto out = at::empty_like(qc);
  auto probs = at::empty({BH, S, S}, q.options());
  const bool drop = p > 0.0;
  auto mask = at::empty({drop ? BH : 0, S, S}, q.options().dtype(at::kByte));
  const int lds_bytes = (2 * S * (hd + 1) + 4 * ATTN_MAX_S) * sizeof(float);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (drop) {
    hipLaunchKernelGGL(attn_fwd_kernel<true>, dim3(BH), dim3(256), lds_bytes,
                       stream, qc.data_ptr<float>(), kc.data_ptr<float>(),
                       vc.data_ptr<float>(), out.data_ptr<float>(),
                       probs.data_ptr<float>(), mask.data_ptr<unsigned char>(),
                       S, hd, (float)scale, (float)p, (float)(1.0 / (1.0 - p)),
                       (uint64_t)seed,
                       offset.has_value() ? offset->data_ptr<long>() : nullptr);
  } else {
    hipLaunchKernelGGL(attn_fwd_kernel<false>, dim3(BH), dim3(256), lds_bytes,
                       stream, qc.data_ptr<float>(), kc.data_ptr<float>(),
                       vc.data_ptr<float>(), out.data_ptr<float>(),
                       probs.data_ptr<float>(), nullptr, S, hd, (float)scale,
                       0.f, 1.f, 0, nullptr);
  }
  return {out, probs, mask};
}

}  // namespace slk
