// Winograd F(2x2, 3x3) convolution for stride-1 3x3 convs (fp32).
// Op sites replaced: every 3x3 s1 Conv2d of the partitioned model zoo
// (reference src/model/VGG16_CIFAR10.py:10-94,
// other/Vanilla_SL/src/model/MobileNetv1_CIFAR10.py:23-167).
//
// y = A^T [ (G g G^T) .* (B^T d B) ] A   per 2x2 output tile (4x4 input
// patch, overlap 2).  2.25x fewer MACs than direct: the elementwise product
// becomes 16 frequency-batched GEMMs  M[f] = U[f][Co,Ci] @ V[f][Ci,T]
// (T = B * OH/2 * OW/2) which run on the existing MFMA tile framework as ONE
// batched launch.  Unfused (transforms round-trip V/M through HBM), so it
// wins where the GEMM is compute-dominated — the high-Ci tail of VGG16 —
// and loses to the direct implicit-GEMM kernel on the big-spatial low-Ci
// layers where the 4x data inflation of V/M eats the MAC savings
// (measured routing in conv2d.hip; MIOpen's advantage on those layers is a
// FUSED Winograd with in-kernel transforms).
//
// Backward-data reuses the whole machinery: gx = winograd-conv of gy with
// the 180-degree-rotated, Co/Ci-transposed weights at pad' = KH-1-p.
// Backward-weight has its own frequency form (dU[f] = (A gy A^T)[f] @
// V_x[f]^T, gw = G^T dU G) in both a transform+GEMM pipeline and a fully
// FUSED kernel (wino_bwdw_fused_kernel).  The fully fused forward
// (wino_fused_kernel) keeps the transforms in-kernel; routing between the
// three forms is measured per shape (conv2d.hip).
#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace slk {

// weight transform U[f] = (G g G^T)[f] over the weight's true dims
// (Cw0, Cw1) = (w.size(0), w.size(1)); FLIP writes U[f][Cw1][Cw0] with the
// taps rotated 180 degrees (the bwd-data weight), else U[f][Cw0][Cw1]
template <bool FLIP>
__global__ void wino_wt_kernel(const float* __restrict__ w,
                               float* __restrict__ U, int Cw0, int Cw1) {
  // thread per (channel pair, frequency ROW): 4x the threads of the obvious
  // per-pair mapping — the small conv weights (64x64: 4096 pairs) gave a
  // 16-BLOCK grid on a 256-CU chip and the launch ran latency-bound at 7-9
  // us; the 4x re-read of w is noise next to that
  const long total = (long)Cw0 * Cw1;
  const long tstride = (long)gridDim.x * blockDim.x;
  for (long e = (long)blockIdx.x * blockDim.x + threadIdx.x; e < 4 * total;
       e += tstride) {
    const int a = (int)(e & 3);          // frequency row
    const long i = e >> 2;
    const int cols = FLIP ? Cw0 : Cw1;
    const int c = (int)(i % cols);
    const int r = (int)(i / cols);
    const int co = FLIP ? c : r;   // index into w's dim 0
    const int ci = FLIP ? r : c;   // index into w's dim 1
    // row a of G g (G = [[1,0,0],[.5,.5,.5],[.5,-.5,.5],[0,0,1]])
    float t[3];
    #pragma unroll
    for (int b = 0; b < 3; ++b) {
      const float g0 = w[(((long)co * Cw1 + ci) * 3 + (FLIP ? 2 : 0)) * 3 +
                         (FLIP ? 2 - b : b)];
      const float g1 = w[(((long)co * Cw1 + ci) * 3 + 1) * 3 +
                         (FLIP ? 2 - b : b)];
      const float g2 = w[(((long)co * Cw1 + ci) * 3 + (FLIP ? 0 : 2)) * 3 +
                         (FLIP ? 2 - b : b)];
      t[b] = a == 0 ? g0
           : a == 1 ? 0.5f * (g0 + g1 + g2)
           : a == 2 ? 0.5f * (g0 - g1 + g2)
           : g2;
    }
    U[((long)(a * 4 + 0) * total) + i] = t[0];
    U[((long)(a * 4 + 1) * total) + i] = 0.5f * (t[0] + t[1] + t[2]);
    U[((long)(a * 4 + 2) * total) + i] = 0.5f * (t[0] - t[1] + t[2]);
    U[((long)(a * 4 + 3) * total) + i] = t[2];
  }
}

// input transform V[f][Ci][T], T = B*tH*tW, tile t = (b, th, tw); the 4x4
// patch starts at (th*2 - pad, tw*2 - pad), zero-padded at the borders
__global__ void wino_in_kernel(const float* __restrict__ x,
                               float* __restrict__ V, int B, int Ci, int H,
                               int W, int tH, int tW, int pad, FastDiv d_T,
                               FastDiv d_thw, FastDiv d_tw) {
  const int T = B * tH * tW;
  const long total = (long)Ci * T;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const unsigned cc = d_T.div((unsigned)i);
    const unsigned t = d_T.mod((unsigned)i, cc);
    const unsigned b = d_thw.div(t);
    const unsigned rem = d_thw.mod(t, b);
    const unsigned th = d_tw.div(rem);
    const unsigned tw = d_tw.mod(rem, th);
    const int ih0 = (int)th * 2 - pad;
    const int iw0 = (int)tw * 2 - pad;
    const float* xp = x + ((long)b * Ci + cc) * H * W;
    float d[4][4];
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      const int ih = ih0 + a;
      const bool hv = (unsigned)ih < (unsigned)H;
      #pragma unroll
      for (int bb = 0; bb < 4; ++bb) {
        const int iw = iw0 + bb;
        const bool v = hv && (unsigned)iw < (unsigned)W;
        d[a][bb] = v ? xp[(long)ih * W + iw] : 0.f;
      }
    }
    // B^T d: rows [d0-d2, d1+d2, d2-d1, d1-d3]
    float u[4][4];
    #pragma unroll
    for (int bb = 0; bb < 4; ++bb) {
      u[0][bb] = d[0][bb] - d[2][bb];
      u[1][bb] = d[1][bb] + d[2][bb];
      u[2][bb] = d[2][bb] - d[1][bb];
      u[3][bb] = d[1][bb] - d[3][bb];
    }
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      const float v0 = u[a][0] - u[a][2];
      const float v1 = u[a][1] + u[a][2];
      const float v2 = u[a][2] - u[a][1];
      const float v3 = u[a][1] - u[a][3];
      V[((long)(a * 4 + 0) * Ci + cc) * T + t] = v0;
      V[((long)(a * 4 + 1) * Ci + cc) * T + t] = v1;
      V[((long)(a * 4 + 2) * Ci + cc) * T + t] = v2;
      V[((long)(a * 4 + 3) * Ci + cc) * T + t] = v3;
    }
  }
}

// output transform y[b][co][2th..][2tw..] = A^T M A (+bias)
__global__ void wino_out_kernel(const float* __restrict__ Mm,
                                float* __restrict__ y,
                                const float* __restrict__ bias, int B, int Co,
                                int OH, int OW, int tH, int tW, FastDiv d_T,
                                FastDiv d_thw, FastDiv d_tw) {
  const int T = B * tH * tW;
  const long total = (long)Co * T;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const unsigned co = d_T.div((unsigned)i);
    const unsigned t = d_T.mod((unsigned)i, co);
    const unsigned b = d_thw.div(t);
    const unsigned rem = d_thw.mod(t, b);
    const unsigned th = d_tw.div(rem);
    const unsigned tw = d_tw.mod(rem, th);
    float m[4][4];
    #pragma unroll
    for (int f = 0; f < 16; ++f)
      m[f >> 2][f & 3] = Mm[((long)f * Co + co) * T + t];
    // A^T m: [m0+m1+m2, m1-m2-m3]
    float u[2][4];
    #pragma unroll
    for (int bb = 0; bb < 4; ++bb) {
      u[0][bb] = m[0][bb] + m[1][bb] + m[2][bb];
      u[1][bb] = m[1][bb] - m[2][bb] - m[3][bb];
    }
    const float bv = bias != nullptr ? bias[co] : 0.f;
    float* yp = y + ((long)b * Co + co) * OH * OW + (long)th * 2 * OW + tw * 2;
    yp[0] = u[0][0] + u[0][1] + u[0][2] + bv;
    yp[1] = u[0][1] - u[0][2] - u[0][3] + bv;
    yp[OW] = u[1][0] + u[1][1] + u[1][2] + bv;
    yp[OW + 1] = u[1][1] - u[1][2] - u[1][3] + bv;
  }
}

// ---- backward-weight --------------------------------------------------
// dY arrives per 2x2 output tile; dM = A dY A^T lifts it to the 4x4
// frequency domain (A = [[1,0],[1,1],[1,-1],[0,-1]]), then
// dU[f][Co][Ci] = dM[f][Co,T] @ V_x[f][Ci,T]^T (16 frequency GEMMs over the
// tile axis — a far better GEMM shape than the direct gather's
// [Co, Ci*9, B*OH*OW] tall-K formulation), and gw = G^T dU G.
__global__ void wino_gyA_kernel(const float* __restrict__ gy,
                                float* __restrict__ Wg, int B, int Co, int OH,
                                int OW, int tH, int tW, FastDiv d_T,
                                FastDiv d_thw, FastDiv d_tw) {
  const int T = B * tH * tW;
  const long total = (long)Co * T;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const unsigned co = d_T.div((unsigned)i);
    const unsigned t = d_T.mod((unsigned)i, co);
    const unsigned b = d_thw.div(t);
    const unsigned rem = d_thw.mod(t, b);
    const unsigned th = d_tw.div(rem);
    const unsigned tw = d_tw.mod(rem, th);
    const float* gp = gy + ((long)b * Co + co) * OH * OW
                      + (long)th * 2 * OW + tw * 2;
    const float g00 = gp[0], g01 = gp[1], g10 = gp[OW], g11 = gp[OW + 1];
    // rows of A gy (4x2): [g0j], [g0j+g1j], [g0j-g1j], [-g1j]
    float r0c0 = g00, r0c1 = g01;
    float r1c0 = g00 + g10, r1c1 = g01 + g11;
    float r2c0 = g00 - g10, r2c1 = g01 - g11;
    float r3c0 = -g10, r3c1 = -g11;
    // columns: dM[a][.] = [c0], [c0+c1], [c0-c1], [-c1]
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      const float c0 = a == 0 ? r0c0 : a == 1 ? r1c0 : a == 2 ? r2c0 : r3c0;
      const float c1 = a == 0 ? r0c1 : a == 1 ? r1c1 : a == 2 ? r2c1 : r3c1;
      Wg[((long)(a * 4 + 0) * Co + co) * T + t] = c0;
      Wg[((long)(a * 4 + 1) * Co + co) * T + t] = c0 + c1;
      Wg[((long)(a * 4 + 2) * Co + co) * T + t] = c0 - c1;
      Wg[((long)(a * 4 + 3) * Co + co) * T + t] = -c1;
    }
  }
}

__global__ void wino_gw_kernel(const float* __restrict__ dU,
                               float* __restrict__ gw, int Co, int Ci) {
  const long total = (long)Co * Ci;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    float u[4][4];
    #pragma unroll
    for (int f = 0; f < 16; ++f) u[f >> 2][f & 3] = dU[(long)f * total + i];
    // t = G^T u (3x4), G = [[1,0,0],[.5,.5,.5],[.5,-.5,.5],[0,0,1]]
    float t[3][4];
    #pragma unroll
    for (int b = 0; b < 4; ++b) {
      t[0][b] = u[0][b] + 0.5f * (u[1][b] + u[2][b]);
      t[1][b] = 0.5f * (u[1][b] - u[2][b]);
      t[2][b] = 0.5f * (u[1][b] + u[2][b]) + u[3][b];
    }
    #pragma unroll
    for (int r = 0; r < 3; ++r) {
      gw[i * 9 + r * 3 + 0] = t[r][0] + 0.5f * (t[r][1] + t[r][2]);
      gw[i * 9 + r * 3 + 1] = 0.5f * (t[r][1] - t[r][2]);
      gw[i * 9 + r * 3 + 2] = 0.5f * (t[r][1] + t[r][2]) + t[r][3];
    }
  }
}

// host entry: 3x3 stride-1 conv via F(2x2,3x3); flip=true computes the
// bwd-data conv (weights rotated + Co/Ci transposed, pad' = 2 - pad)
at::Tensor matmul_f32(const at::Tensor&, const at::Tensor&, bool, bool,
                      c10::optional<at::Tensor>, bool);

at::Tensor conv2d_wino(const at::Tensor& x, const at::Tensor& w,
                       c10::optional<at::Tensor> bias, int pad, bool flip) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.scalar_type() == at::kFloat);
  TORCH_CHECK(w.size(2) == 3 && w.size(3) == 3, "wino: 3x3 only");
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  const int B = x.size(0), H = x.size(2), W = x.size(3);
  const int Co = flip ? w.size(1) : w.size(0);
  const int Ci = flip ? w.size(0) : w.size(1);
  TORCH_CHECK(x.size(1) == Ci, "wino: channel mismatch");
  const int OH = H + 2 * pad - 2, OW = W + 2 * pad - 2;
  TORCH_CHECK(OH % 2 == 0 && OW % 2 == 0, "wino: even output dims only");
  const int tH = OH / 2, tW = OW / 2;
  const int T = B * tH * tW;
  auto stream = c10::hip::getCurrentHIPStream().stream();

  auto U = at::empty({16, Co, Ci}, x.options());
  auto V = at::empty({16, Ci, T}, x.options());
  {
    const long tot = (long)Co * Ci * 4;   // thread per (pair, freq row)
    const int grid = (int)std::min<long>((tot + 255) / 256, 4096);
    // always the WEIGHT's own dims; FLIP handles the transpose internally
    if (flip) {
      hipLaunchKernelGGL(wino_wt_kernel<true>, dim3(grid), dim3(256), 0,
                         stream, wc.data_ptr<float>(), U.data_ptr<float>(),
                         (int)w.size(0), (int)w.size(1));
    } else {
      hipLaunchKernelGGL(wino_wt_kernel<false>, dim3(grid), dim3(256), 0,
                         stream, wc.data_ptr<float>(), U.data_ptr<float>(),
                         (int)w.size(0), (int)w.size(1));
    }
  }
  {
    FastDiv d_T, d_thw, d_tw;
    d_T.init(T);
    d_thw.init(tH * tW);
    d_tw.init(tW);
    const long tot = (long)Ci * T;
    const int grid = (int)std::min<long>((tot + 255) / 256, 8192);
    hipLaunchKernelGGL(wino_in_kernel, dim3(grid), dim3(256), 0, stream,
                       xc.data_ptr<float>(), V.data_ptr<float>(), B, Ci, H, W,
                       tH, tW, pad, d_T, d_thw, d_tw);
  }
  // M[f] = U[f] @ V[f]: one batched MFMA GEMM launch (batch = 16)
  auto Mm = matmul_f32(U, V, false, false, c10::nullopt, false);
  auto y = at::empty({B, Co, OH, OW}, x.options());
  {
    FastDiv d_T, d_thw, d_tw;
    d_T.init(T);
    d_thw.init(tH * tW);
    d_tw.init(tW);
    const long tot = (long)Co * T;
    const int grid = (int)std::min<long>((tot + 255) / 256, 8192);
    hipLaunchKernelGGL(wino_out_kernel, dim3(grid), dim3(256), 0, stream,
                       Mm.data_ptr<float>(), y.data_ptr<float>(),
                       bias.has_value() ? bias->data_ptr<float>() : nullptr,
                       B, Co, OH, OW, tH, tW, d_T, d_thw, d_tw);
  }
  return y;
}


// ---- fused F(2x2,3x3) kernel ------------------------------------------
// One kernel: input transform in-register, 16-frequency MFMA accumulation,
// output transform in the epilogue — no V/M HBM round trip (the unfused
// pipeline's 4x data inflation).  A block owns a (32 co x 32 tile) output
// patch across ALL 16 frequencies, so each 4x4 input patch is read once and
// feeds every frequency.  Per wave: 16x16 (co, t) fragment x 16 freq = 16
// f32x4 accumulators (64 AGPR).  K-loop over ci in steps of 8.
// Weights arrive PRE-TRANSFORMED (U[16][Co][Ci], wino_wt_kernel — tiny).
constexpr int WF_CO = 32;   // block co tile
constexpr int WF_T = 32;    // block tile-column tile
constexpr int WF_CK = 8;    // ci step
// LDS row strides padded to dodge the worst bank conflicts
constexpr int WF_ULD = WF_CO + 1;   // U_lds[f][ci][co]
constexpr int WF_VLD = WF_T + 1;    // V_lds[f][ci][t]

// ci_per: the ci-range length per grid.z slice.  The output transform is
// LINEAR in M, so slices transform their PARTIAL sums and atomicAdd into a
// zeroed y — this fills the chip for small-T/high-Ci shapes whose natural
// grid is far below one block per CU.  Slice 0 adds the bias.
template <bool FLIP>
__global__ __launch_bounds__(256, 3) void wino_fused_kernel(
    const float* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, float* __restrict__ y, int B, int Ci,
    int H, int W, int Co, int OH, int OW, int tH, int tW, int pad,
    int ci_per, int CW, FastDiv d_thw, FastDiv d_tw) {
  const int ci_begin = blockIdx.z * ci_per;
  const int ci_end = min(Ci, ci_begin + ci_per);
  const bool accumulate = gridDim.z > 1;
  // single-buffer two-barrier loop: double buffering would cost 67.6 KB of
  // LDS (2 blocks/CU); at 33.8 KB four blocks fit and the long 32-MFMA
  // phase gives co-resident waves the latency cover instead
  __shared__ float Ul[16 * WF_CK * WF_ULD];
  __shared__ float Vl[16 * WF_CK * WF_VLD];

  const int T = B * tH * tW;
  const int co0 = blockIdx.y * WF_CO;
  const int t0 = blockIdx.x * WF_T;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wco = (wid >> 1) * 16;   // wave offset in the 32-co tile
  const int wt = (wid & 1) * 16;     // wave offset in the 32-t tile
  const int frag_r = lane >> 4;      // k sub-index (0..3)
  const int frag_c = lane & 15;

  f32x4 acc[16];
  #pragma unroll
  for (int f = 0; f < 16; ++f) acc[f] = f32x4{};

  // staging assignments -----------------------------------------------
  // V: thread -> one (ci, t) pair: ci = tid >> 5 (8), t = tid & 31 (32)
  const int v_ci = tid >> 5;
  const int v_t = tid & 31;
  const int tg = t0 + v_t;
  const unsigned vb = d_thw.div((unsigned)tg);
  const unsigned vrem = d_thw.mod((unsigned)tg, vb);
  const unsigned vth = d_tw.div(vrem);
  const unsigned vtw = d_tw.mod(vrem, vth);
  const int ih0 = (int)vth * 2 - pad;
  const int iw0 = (int)vtw * 2 - pad;
  const long xplane = ((long)vb * Ci + v_ci) * H * W;
  // weights: thread -> one (co, ci) pair (32co x 8ci = 256); the 3x3 taps
  // load raw and transform IN-KERNEL (U = G w G^T, the exact wino_wt math) —
  // the separate wt launch and the 16*Co*Ci frequency tensor's HBM
  // write+read round trip disappear.  FLIP (bwd-data) swaps the channel
  // roles and rotates the taps 180 deg ((2-r, 2-s) == tap 8-k).
  const int w_co = tid >> 3;
  const int w_ci = tid & 7;

  // software pipeline: next step's 4x4 patch + 9 weight taps load into
  // registers UNDER the (long, 32-MFMA) compute phase; the transforms + LDS
  // stores run at the top of the next iteration
  float d[4][4];
  float wr[9];

  auto load_regs = [&](int ci0) {
    const float* xp = x + xplane + (long)ci0 * H * W;
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      const int ih = ih0 + a;
      const bool hv = (unsigned)ih < (unsigned)H;
      #pragma unroll
      for (int bb = 0; bb < 4; ++bb) {
        const int iw = iw0 + bb;
        const bool v = hv && (unsigned)iw < (unsigned)W;
        d[a][bb] = v ? xp[(long)ih * W + iw] : 0.f;
      }
    }
    const long wbase = FLIP
        ? ((long)(ci0 + w_ci) * CW + (co0 + w_co)) * 9
        : ((long)(co0 + w_co) * CW + (ci0 + w_ci)) * 9;
    #pragma unroll
    for (int k = 0; k < 9; ++k) wr[k] = w[wbase + (FLIP ? 8 - k : k)];
  };

  auto store_stage = [&]() {
    float u[4][4];
    #pragma unroll
    for (int bb = 0; bb < 4; ++bb) {
      u[0][bb] = d[0][bb] - d[2][bb];
      u[1][bb] = d[1][bb] + d[2][bb];
      u[2][bb] = d[2][bb] - d[1][bb];
      u[3][bb] = d[1][bb] - d[3][bb];
    }
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      Vl[((a * 4 + 0) * WF_CK + v_ci) * WF_VLD + v_t] = u[a][0] - u[a][2];
      Vl[((a * 4 + 1) * WF_CK + v_ci) * WF_VLD + v_t] = u[a][1] + u[a][2];
      Vl[((a * 4 + 2) * WF_CK + v_ci) * WF_VLD + v_t] = u[a][2] - u[a][1];
      Vl[((a * 4 + 3) * WF_CK + v_ci) * WF_VLD + v_t] = u[a][1] - u[a][3];
    }
    // U = G w G^T for this thread's (co, ci) pair — same op order as
    // wino_wt_kernel, so the result is bit-identical to the staged path
    float t0[3], t1[3], t2[3], t3[3];
    #pragma unroll
    for (int bb = 0; bb < 3; ++bb) {
      const float g0 = wr[bb], g1 = wr[3 + bb], g2 = wr[6 + bb];
      t0[bb] = g0;
      t1[bb] = 0.5f * (g0 + g1 + g2);
      t2[bb] = 0.5f * (g0 - g1 + g2);
      t3[bb] = g2;
    }
    const float* tr[4] = {t0, t1, t2, t3};
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      const float* t = tr[a];
      Ul[((a * 4 + 0) * WF_CK + w_ci) * WF_ULD + w_co] = t[0];
      Ul[((a * 4 + 1) * WF_CK + w_ci) * WF_ULD + w_co] = 0.5f * (t[0] + t[1] + t[2]);
      Ul[((a * 4 + 2) * WF_CK + w_ci) * WF_ULD + w_co] = 0.5f * (t[0] - t[1] + t[2]);
      Ul[((a * 4 + 3) * WF_CK + w_ci) * WF_ULD + w_co] = t[2];
    }
  };

  load_regs(ci_begin);
  for (int ci0 = ci_begin; ci0 < ci_end; ci0 += WF_CK) {
    store_stage();
    __syncthreads();
    if (ci0 + WF_CK < ci_end) load_regs(ci0 + WF_CK);
    #pragma unroll
    for (int f = 0; f < 16; ++f) {
      #pragma unroll
      for (int kk = 0; kk < WF_CK / 4; ++kk) {
        const int kr = kk * 4 + frag_r;
        const float a = Ul[(f * WF_CK + kr) * WF_ULD + wco + frag_c];
        const float b = Vl[(f * WF_CK + kr) * WF_VLD + wt + frag_c];
        acc[f] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: lane holds, per freq, rows co = wco + frag_r*4 + i (i<4) at
  // col t = wt + frag_c -> all 16 freqs of each (co, t): y = A^T M A + bias
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int co = co0 + wco + frag_r * 4 + i;
    const int t = t0 + wt + frag_c;
    if (co >= Co || t >= T) continue;
    float m[4][4];
    #pragma unroll
    for (int f = 0; f < 16; ++f) m[f >> 2][f & 3] = acc[f][i];
    float u0[4], u1[4];
    #pragma unroll
    for (int bb = 0; bb < 4; ++bb) {
      u0[bb] = m[0][bb] + m[1][bb] + m[2][bb];
      u1[bb] = m[1][bb] - m[2][bb] - m[3][bb];
    }
    const float bv = (bias != nullptr && blockIdx.z == 0) ? bias[co] : 0.f;
    const unsigned b = d_thw.div((unsigned)t);
    const unsigned rem = d_thw.mod((unsigned)t, b);
    const unsigned th = d_tw.div(rem);
    const unsigned tw = d_tw.mod(rem, th);
    float* yp = y + ((long)b * Co + co) * OH * OW + (long)th * 2 * OW + tw * 2;
    const float o00 = u0[0] + u0[1] + u0[2] + bv;
    const float o01 = u0[1] - u0[2] - u0[3] + bv;
    const float o10 = u1[0] + u1[1] + u1[2] + bv;
    const float o11 = u1[1] - u1[2] - u1[3] + bv;
    if (accumulate) {
      atomicAdd(&yp[0], o00);
      atomicAdd(&yp[1], o01);
      atomicAdd(&yp[OW], o10);
      atomicAdd(&yp[OW + 1], o11);
    } else {
      yp[0] = o00;
      yp[1] = o01;
      yp[OW] = o10;
      yp[OW + 1] = o11;
    }
  }
}

at::Tensor conv2d_wino_bwdw(const at::Tensor& gy, const at::Tensor& x,
                            int pad) {
  TORCH_CHECK(gy.is_cuda() && x.is_cuda() && gy.dim() == 4 && x.dim() == 4);
  auto gyc = gy.contiguous();
  auto xc = x.contiguous();
  const int B = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int Co = gy.size(1), OH = gy.size(2), OW = gy.size(3);
  TORCH_CHECK(OH == H + 2 * pad - 2 && OW == W + 2 * pad - 2,
              "wino_bwdw geometry");
  TORCH_CHECK(OH % 2 == 0 && OW % 2 == 0, "wino_bwdw: even output dims only");
  const int tH = OH / 2, tW = OW / 2;
  const int T = B * tH * tW;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  FastDiv d_T, d_thw, d_tw;
  d_T.init(T);
  d_thw.init(tH * tW);
  d_tw.init(tW);

  auto Wg = at::empty({16, Co, T}, x.options());
  auto Vx = at::empty({16, Ci, T}, x.options());
  {
    const long tot = (long)Co * T;
    const int grid = (int)std::min<long>((tot + 255) / 256, 8192);
    hipLaunchKernelGGL(wino_gyA_kernel, dim3(grid), dim3(256), 0, stream,
                       gyc.data_ptr<float>(), Wg.data_ptr<float>(), B, Co, OH,
                       OW, tH, tW, d_T, d_thw, d_tw);
  }
  {
    const long tot = (long)Ci * T;
    const int grid = (int)std::min<long>((tot + 255) / 256, 8192);
    hipLaunchKernelGGL(wino_in_kernel, dim3(grid), dim3(256), 0, stream,
                       xc.data_ptr<float>(), Vx.data_ptr<float>(), B, Ci, H, W,
                       tH, tW, pad, d_T, d_thw, d_tw);
  }
  // dU[f][Co][Ci] = Wg[f] @ Vx[f]^T : 16 frequency GEMMs, K = tiles
  auto dU = matmul_f32(Wg, Vx, false, true, c10::nullopt, false);
  auto gw = at::empty({Co, Ci, 3, 3}, x.options());
  {
    const long tot = (long)Co * Ci;
    const int grid = (int)std::min<long>((tot + 255) / 256, 4096);
    hipLaunchKernelGGL(wino_gw_kernel, dim3(grid), dim3(256), 0, stream,
                       dU.data_ptr<float>(), gw.data_ptr<float>(), Co, Ci);
  }
  return gw;
}

// ---- fused backward-weight --------------------------------------------
// Twin of wino_fused_kernel with the TILE axis as the reduction: block owns
// a (32co x 32ci) patch of gw across all 16 frequencies, K-loop over tiles
// staging (A gy A^T) and (B^T x B) patches transformed in-register, epilogue
// applies G^T dU G.  grid.z slices the tile range; slices write per-split
// slabs (non-atomic) reduced by slab_reduce (the 2.3M-atomicAdd alternative
// re-creates the split-K atomic pathology measured in round 2).
__global__ __launch_bounds__(256, 2) void wino_bwdw_fused_kernel(
    const float* __restrict__ gy, const float* __restrict__ x,
    float* __restrict__ slab, int B, int Ci, int H, int W, int Co, int OH,
    int OW, int tH, int tW, int pad, int t_per, FastDiv d_thw, FastDiv d_tw) {
  __shared__ float Gl[16 * 8 * 33];   // [f][tloc][co]
  __shared__ float Xl[16 * 8 * 33];   // [f][tloc][ci]

  const int T = B * tH * tW;
  const int ci0 = blockIdx.y * 32;
  const int co0 = blockIdx.x * 32;
  const int t_begin = blockIdx.z * t_per;
  const int t_end = min(T, t_begin + t_per);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wco = (wid >> 1) * 16;
  const int wci = (wid & 1) * 16;
  const int frag_r = lane >> 4;
  const int frag_c = lane & 15;

  f32x4 acc[16];
  #pragma unroll
  for (int f = 0; f < 16; ++f) acc[f] = f32x4{};

  // staging: thread -> (channel = tid>>3, tloc = tid&7) for BOTH sides
  const int s_ch = tid >> 3;          // 0..31
  const int s_t = tid & 7;            // 0..7

  float gg[2][2];     // gy 2x2 tile (for co0 + s_ch)
  float xd[4][4];     // x 4x4 patch (for ci0 + s_ch)

  auto load_regs = [&](int t0) {
    const int t = t0 + s_t;
    const int tc = t < T ? t : T - 1;
    const bool tv = t < t_end;
    const unsigned b = d_thw.div((unsigned)tc);
    const unsigned rem = d_thw.mod((unsigned)tc, b);
    const unsigned th = d_tw.div(rem);
    const unsigned tw = d_tw.mod(rem, th);
    {
      const float* gp = gy + ((long)b * Co + co0 + s_ch) * OH * OW
                        + (long)th * 2 * OW + tw * 2;
      gg[0][0] = tv ? gp[0] : 0.f;
      gg[0][1] = tv ? gp[1] : 0.f;
      gg[1][0] = tv ? gp[OW] : 0.f;
      gg[1][1] = tv ? gp[OW + 1] : 0.f;
    }
    {
      const int ih0 = (int)th * 2 - pad;
      const int iw0 = (int)tw * 2 - pad;
      const float* xp = x + ((long)b * Ci + ci0 + s_ch) * H * W;
      #pragma unroll
      for (int a = 0; a < 4; ++a) {
        const int ih = ih0 + a;
        const bool hv = tv && (unsigned)ih < (unsigned)H;
        #pragma unroll
        for (int bb = 0; bb < 4; ++bb) {
          const int iw = iw0 + bb;
          const bool v = hv && (unsigned)iw < (unsigned)W;
          xd[a][bb] = v ? xp[(long)ih * W + iw] : 0.f;
        }
      }
    }
  };

  auto store_stage = [&]() {
    // dM = A gy A^T (A = [[1,0],[1,1],[1,-1],[0,-1]])
    const float r0c0 = gg[0][0], r0c1 = gg[0][1];
    const float r1c0 = gg[0][0] + gg[1][0], r1c1 = gg[0][1] + gg[1][1];
    const float r2c0 = gg[0][0] - gg[1][0], r2c1 = gg[0][1] - gg[1][1];
    const float r3c0 = -gg[1][0], r3c1 = -gg[1][1];
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      const float c0 = a == 0 ? r0c0 : a == 1 ? r1c0 : a == 2 ? r2c0 : r3c0;
      const float c1 = a == 0 ? r0c1 : a == 1 ? r1c1 : a == 2 ? r2c1 : r3c1;
      Gl[((a * 4 + 0) * 8 + s_t) * 33 + s_ch] = c0;
      Gl[((a * 4 + 1) * 8 + s_t) * 33 + s_ch] = c0 + c1;
      Gl[((a * 4 + 2) * 8 + s_t) * 33 + s_ch] = c0 - c1;
      Gl[((a * 4 + 3) * 8 + s_t) * 33 + s_ch] = -c1;
    }
    // B^T d B
    float u[4][4];
    #pragma unroll
    for (int bb = 0; bb < 4; ++bb) {
      u[0][bb] = xd[0][bb] - xd[2][bb];
      u[1][bb] = xd[1][bb] + xd[2][bb];
      u[2][bb] = xd[2][bb] - xd[1][bb];
      u[3][bb] = xd[1][bb] - xd[3][bb];
    }
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      Xl[((a * 4 + 0) * 8 + s_t) * 33 + s_ch] = u[a][0] - u[a][2];
      Xl[((a * 4 + 1) * 8 + s_t) * 33 + s_ch] = u[a][1] + u[a][2];
      Xl[((a * 4 + 2) * 8 + s_t) * 33 + s_ch] = u[a][2] - u[a][1];
      Xl[((a * 4 + 3) * 8 + s_t) * 33 + s_ch] = u[a][1] - u[a][3];
    }
  };

  load_regs(t_begin);
  for (int t0 = t_begin; t0 < t_end; t0 += 8) {
    store_stage();
    __syncthreads();
    if (t0 + 8 < t_end) load_regs(t0 + 8);
    #pragma unroll
    for (int f = 0; f < 16; ++f) {
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        const int kr = kk * 4 + frag_r;
        const float a = Gl[(f * 8 + kr) * 33 + wco + frag_c];
        const float b = Xl[(f * 8 + kr) * 33 + wci + frag_c];
        acc[f] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: lane holds per freq rows co = wco + frag_r*4 + i, col ci =
  // wci + frag_c; gw = G^T dU G per (co, ci), written to this slice's slab
  const long numel = (long)Co * Ci * 9;
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int co = co0 + wco + frag_r * 4 + i;
    const int ci = ci0 + wci + frag_c;
    if (co >= Co || ci >= Ci) continue;
    float u[4][4];
    #pragma unroll
    for (int f = 0; f < 16; ++f) u[f >> 2][f & 3] = acc[f][i];
    float t3[3][4];
    #pragma unroll
    for (int bb = 0; bb < 4; ++bb) {
      t3[0][bb] = u[0][bb] + 0.5f * (u[1][bb] + u[2][bb]);
      t3[1][bb] = 0.5f * (u[1][bb] - u[2][bb]);
      t3[2][bb] = 0.5f * (u[1][bb] + u[2][bb]) + u[3][bb];
    }
    float* gp = slab + (long)blockIdx.z * numel + ((long)co * Ci + ci) * 9;
    #pragma unroll
    for (int r = 0; r < 3; ++r) {
      gp[r * 3 + 0] = t3[r][0] + 0.5f * (t3[r][1] + t3[r][2]);
      gp[r * 3 + 1] = 0.5f * (t3[r][1] - t3[r][2]);
      gp[r * 3 + 2] = 0.5f * (t3[r][1] + t3[r][2]) + t3[r][3];
    }
  }
}

__global__ void wino_slab_reduce_kernel(const float* __restrict__ slab,
                                        int splits, float* __restrict__ out,
                                        long numel) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < numel;
       i += stride) {
    float s = 0.f;
    for (int k = 0; k < splits; ++k) s += slab[(long)k * numel + i];
    out[i] = s;
  }
}

at::Tensor conv2d_wino_bwdw_fused(const at::Tensor& gy, const at::Tensor& x,
                                  int pad) {
  TORCH_CHECK(gy.is_cuda() && x.is_cuda() && gy.dim() == 4 && x.dim() == 4);
  auto gyc = gy.contiguous();
  auto xc = x.contiguous();
  const int B = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int Co = gy.size(1), OH = gy.size(2), OW = gy.size(3);
  TORCH_CHECK(OH == H + 2 * pad - 2 && OW == W + 2 * pad - 2,
              "wino_bwdw_fused geometry");
  TORCH_CHECK(OH % 2 == 0 && OW % 2 == 0 && Co % 32 == 0 && Ci % 32 == 0,
              "wino_bwdw_fused shape requirements");
  const int tH = OH / 2, tW = OW / 2;
  const int T = B * tH * tW;
  auto stream = c10::hip::getCurrentHIPStream().stream();

  const long base_blocks = (long)(Co / 32) * (Ci / 32);
  int splits = 1;
  if (base_blocks < 256) {
    splits = (int)((256 + base_blocks - 1) / base_blocks);
    const int max_splits = (T + 7) / 8;
    if (splits > max_splits) splits = max_splits;
  }
  int t_per = ((T / splits + 7) / 8) * 8;
  if (t_per < 8) t_per = 8;
  splits = (T + t_per - 1) / t_per;

  auto slab = at::empty({splits, Co, Ci, 3, 3}, x.options());
  FastDiv d_thw, d_tw;
  d_thw.init(tH * tW);
  d_tw.init(tW);
  dim3 grid(Co / 32, Ci / 32, splits);
  hipLaunchKernelGGL(wino_bwdw_fused_kernel, grid, dim3(256), 0, stream,
                     gyc.data_ptr<float>(), xc.data_ptr<float>(),
                     slab.data_ptr<float>(), B, Ci, H, W, Co, OH, OW, tH, tW,
                     pad, t_per, d_thw, d_tw);
  auto gw = at::empty({Co, Ci, 3, 3}, x.options());
  if (splits == 1) {
    return slab.view({Co, Ci, 3, 3});
  }
  const long numel = (long)Co * Ci * 9;
  const int rgrid = (int)std::min<long>((numel + 255) / 256, 4096);
  hipLaunchKernelGGL(wino_slab_reduce_kernel, dim3(rgrid), dim3(256), 0,
                     stream, slab.data_ptr<float>(), splits,
                     gw.data_ptr<float>(), numel);
  return gw;
}

// fully-fused variant: transforms in-kernel, no V/M round trip.  Shape
// requirements: Co % 32 == 0, T % 32 == 0, Ci % 8 == 0.  SLK_WINO_FUSED=0
// falls back to the unfused pipeline (A/B).
at::Tensor conv2d_wino_fused(const at::Tensor& x, const at::Tensor& w,
                             c10::optional<at::Tensor> bias, int pad,
                             bool flip) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.scalar_type() == at::kFloat);
  TORCH_CHECK(w.size(2) == 3 && w.size(3) == 3, "wino_fused: 3x3 only");
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  const int B = x.size(0), H = x.size(2), W = x.size(3);
  const int Co = flip ? w.size(1) : w.size(0);
  const int Ci = flip ? w.size(0) : w.size(1);
  TORCH_CHECK(x.size(1) == Ci, "wino_fused: channel mismatch");
  const int OH = H + 2 * pad - 2, OW = W + 2 * pad - 2;
  const int tH = OH / 2, tW = OW / 2;
  const int T = B * tH * tW;
  TORCH_CHECK(Co % 32 == 0 && T % 32 == 0 && Ci % 8 == 0 && OH % 2 == 0
              && OW % 2 == 0, "wino_fused shape requirements");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  // the weight transform runs IN-KERNEL (one (co,ci) pair per thread per
  // staging step) — no wt launch, no 16*Co*Ci frequency tensor
  const int CW = (int)w.size(1);
  // ci-split so the grid reaches ~1 block/CU (output transform is linear in
  // M: slices atomicAdd partial y tiles; slice count keeps ci chunks at
  // multiples of the CK=8 step)
  const long base_blocks = (long)(T / 32) * (Co / 32);
  int splits = 1;
  if (base_blocks < 256) {
    splits = (int)((256 + base_blocks - 1) / base_blocks);
    const int max_splits = Ci / WF_CK;
    if (splits > max_splits) splits = max_splits;
  }
  int ci_per = ((Ci / splits + WF_CK - 1) / WF_CK) * WF_CK;
  splits = (Ci + ci_per - 1) / ci_per;

  at::Tensor y;
  if (splits > 1) {
    y = at::empty({B, Co, OH, OW}, x.options());
    slk_zero_async(y.data_ptr<float>(), y.numel(), stream);
  } else {
    y = at::empty({B, Co, OH, OW}, x.options());
  }
  FastDiv d_thw, d_tw;
  d_thw.init(tH * tW);
  d_tw.init(tW);
  dim3 grid(T / 32, Co / 32, splits);
  if (flip) {
    hipLaunchKernelGGL(wino_fused_kernel<true>, grid, dim3(256), 0, stream,
                       xc.data_ptr<float>(), wc.data_ptr<float>(),
                       bias.has_value() ? bias->data_ptr<float>() : nullptr,
                       y.data_ptr<float>(), B, Ci, H, W, Co, OH, OW, tH, tW,
                       pad, ci_per, CW, d_thw, d_tw);
  } else {
    hipLaunchKernelGGL(wino_fused_kernel<false>, grid, dim3(256), 0, stream,
                       xc.data_ptr<float>(), wc.data_ptr<float>(),
                       bias.has_value() ? bias->data_ptr<float>() : nullptr,
                       y.data_ptr<float>(), B, Ci, H, W, Co, OH, OW, tH, tW,
                       pad, ci_per, CW, d_thw, d_tw);
  }
  return y;
}

}  // namespace slk
