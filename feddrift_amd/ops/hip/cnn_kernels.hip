// feddrift MI355X (gfx950 / CDNA4) kernels — CNN_DropOut training + eval.
//
// Hand-written CDNA4 pipeline for the reference's MNIST/FEMNIST CNN
// (fedml_api/model/cv/cnn.py:113-135: 784 -> 28x28 -> conv3x3(32) ->
// conv3x3(64) [NO activation after either conv] -> maxpool2x2 ->
// dropout(.25) -> flatten(9216) -> relu(fc 128) -> dropout(.5) -> fc O ->
// in-graph Softmax; the training loss is CE applied to that softmax output
// — the double-softmax quirk, FedAvgEnsTrainer.py:73 — and the optimizer
// is SGD or Adam(amsgrad, wd), FedAvgEnsTrainer.py:23-33).
//
// Design (not a port — the reference runs this as per-(client, model)
// eager torch with CPU<->GPU model movement every round):
//   * ONE epoch = one fused multi-kernel launch sequence batched over ALL
//     (client, model) pairs of the round; per-pair weights stay resident
//     in flat HBM rows the whole time.
//   * gradient buffers use exclusive-owner writes (each block owns a
//     disjoint grad slice and loops the batch internally) — no grad
//     zeroing pass and no atomics on the wgrad paths.
//   * dropout masks are counter-based hashes of (seed, element id), so the
//     backward pass recomputes them instead of storing them.
//   * the softmax+CE tail computes dL/dz2 in the forward head kernel
//     (one kernel fuses fc2 forward, both softmaxes, and the CE gradient).
//   * fp32 everywhere: dtype parity with the reference's training, and
//     f32-input MFMA on gfx950 is bit-exact against an fmaf chain.
//
// Conv shapes (28x28 in): conv1 -> [32, 26, 26], conv2 -> [64, 24, 24],
// pool -> [64, 12, 12] = 9216 = fc1 input. fc1 -> 128, fc2 -> O (10/62).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <cmath>

#define WG 256

// fixed geometry of CNN_DropOut
#define IN_W 28
#define C1 32
#define S1 26           // conv1 output side
#define C2 64
#define S2 24           // conv2 output side
#define SP 12           // pooled side
#define NF 9216         // C2*SP*SP
#define NH 128
#define X1N (C1 * S1 * S1)   // 21632
#define Z2N (C2 * S2 * S2)   // 36864
#define D_IN 784

// flat parameter offsets (state_dict order; models/generic_packer.py)
#define OFF_W1C 0
#define OFF_B1C 288
#define OFF_W2C 320
#define OFF_B2C 18752
#define OFF_W1F 18816
#define OFF_B1F 1198464
#define OFF_W2F 1198592
// OFF_B2F = OFF_W2F + O*NH (runtime O)

struct CnnArgs {
  float* __restrict__ work;          // [G, P] live weights
  float* __restrict__ grad;          // [G, P]
  const int64_t* __restrict__ rows;  // [G] replica rows (opt state index)
  const float* __restrict__ x;       // arena [N, 784]
  const int64_t* __restrict__ y;     // arena [N]
  const int64_t* __restrict__ step_off;  // [G, E]
  const int64_t* __restrict__ step_len;  // [G, E]
  const float* __restrict__ x_mask;  // [G, 784] or null
  // workspace
  float* __restrict__ x1;            // [G, B, X1N]
  float* __restrict__ a2;            // [G, B, NF]  pooled+dropout1
  unsigned char* __restrict__ pidx;  // [G, B, NF]  pool argmax 0..3
  float* __restrict__ z1;            // [G, B, NH]  fc1 pre-activation
  float* __restrict__ a1;            // [G, B, NH]  relu+dropout2
  float* __restrict__ dz2;           // [G, B, O]   dL/d(fc2 out)
  float* __restrict__ dz1;           // [G, B, NH]
  float* __restrict__ da2;           // [G, B, NF]
  float* __restrict__ zz2;           // [G, B, Z2N] z2 (fwd) / dz2 (bwd),
                                     //   channels-last
  float* __restrict__ dx1;           // [G, B, X1N] channels-last
  float* __restrict__ c1part;        // [G, B, 320] conv1-wgrad partials
  float* __restrict__ wtf;           // [G, 9, 32, 64] conv2 W (fwd)
  float* __restrict__ wtd;           // [G, 9, 64, 32] conv2 W (dgrad)
  float* __restrict__ z1part;        // [G, FC1_KS, B, NH]
  float* __restrict__ w2part;        // [G, 9, W2_KS, 32, 64]
  float* __restrict__ b2part;        // [G, B, 64] conv2 bias partials
  // optimizer state (indexed by rows[g])
  float* __restrict__ m;
  float* __restrict__ v;
  float* __restrict__ vmax;
  int* __restrict__ t;
  const float* __restrict__ lr;      // [n_rows]
  float wd;
  float p1, p2;                      // dropout probs
  unsigned long long seed;           // per (round, epoch)
  long long g0;                      // global pair-index base (mask hash)
  int G, B, E, e, O, P, opt, w2ms;
};

#define OPT_SGD 0
#define OPT_ADAM 1

typedef float f32x4 __attribute__((ext_vector_type(4)));
#define EVAL_BK 32

__device__ __forceinline__ float hash_u01(unsigned long long s,
                                          unsigned long long id) {
  unsigned long long x = s + id * 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  x ^= x >> 31;
  return (float)(x >> 40) * (1.f / 16777216.f);
}

// dropout keep-scale for element id (1/(1-p) when kept, 0 when dropped)
__device__ __forceinline__ float drop_scale(const CnnArgs& a, int which,
                                            long long g, int b,
                                            long long j) {
  const float p = which == 0 ? a.p1 : a.p2;
  if (p <= 0.f) return 1.f;
  const long long nj = which == 0 ? NF : NH;
  const unsigned long long id =
      (unsigned long long)(((a.g0 + g) * (long long)a.E + a.e) *
                               (long long)a.B + b) * (2 * NF) +
      which * nj + j;
  return hash_u01(a.seed, id) >= p ? 1.f / (1.f - p) : 0.f;
}

__device__ __forceinline__ int step_n(const CnnArgs& a, int g) {
  return (int)a.step_len[(long long)g * a.E + a.e];
}
__device__ __forceinline__ long long step_o(const CnnArgs& a, int g) {
  return (long long)a.step_off[(long long)g * a.E + a.e];
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

// conv1 forward, block per (g, b): the masked input and the conv1
// weights stage in LDS once; each thread emits channels-last x1
// elements (9 fused MACs each). x1 is CHANNELS-LAST [G, B, 676, 32] so
// the conv2 MFMA stagers read contiguous ci slices.
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv1_fwd(CnnArgs a) {
  const int g = blockIdx.x / a.B;
  const int b = blockIdx.x - g * a.B;
  if (b >= step_n(a, g)) return;
  const int tid = threadIdx.x;
  __shared__ __attribute__((aligned(16))) float xin[D_IN];
  __shared__ __attribute__((aligned(16))) float wc[288 + C1];
  const float* w = a.work + (long long)g * a.P;
  const float* xs = a.x + (step_o(a, g) + b) * D_IN;
  const float* xm = a.x_mask ? a.x_mask + (long long)g * D_IN : nullptr;
  for (int d = tid; d < D_IN; d += WG)
    xin[d] = xm ? xs[d] * xm[d] : xs[d];
  for (int i = tid; i < 288 + C1; i += WG) wc[i] = w[OFF_W1C + i];
  __syncthreads();
  float* out = a.x1 + ((long long)g * a.B + b) * X1N;
  for (int e = tid; e < X1N; e += WG) {
    const int pp = e / C1;
    const int c = e - pp * C1;
    const int oy = pp / S1, ox = pp - (pp / S1) * S1;
    float z = wc[288 + c];
#pragma unroll
    for (int ky = 0; ky < 3; ++ky)
#pragma unroll
      for (int kx = 0; kx < 3; ++kx)
        z = fmaf(xin[(oy + ky) * IN_W + ox + kx],
                 wc[c * 9 + ky * 3 + kx], z);
    out[e] = z;
  }
}

// reshape conv2 weights for the MFMA stagers (per pair, per epoch):
// wtf[g][kyx][ci][co] (fwd B-operand) and wtd[g][kyx][co][ci] (dgrad)
extern "C" __global__ __launch_bounds__(WG)
void cnn_w2_reshape(CnnArgs a) {
  const long long total = (long long)a.G * C2 * C1 * 9;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / (C2 * C1 * 9));
    const int r = (int)(q - (long long)g * C2 * C1 * 9);
    const int co = r / (C1 * 9);
    const int t = r - co * C1 * 9;
    const int ci = t / 9;
    const int kyx = t - ci * 9;
    const float wv = a.work[(long long)g * a.P + OFF_W2C + r];
    a.wtf[((long long)g * 9 + kyx) * 2048 + ci * C2 + co] = wv;
    a.wtd[((long long)g * 9 + kyx) * 2048 + co * C1 + ci] = wv;
  }
}

// conv2 forward as implicit GEMM on f32 MFMA: per (g, b, ptile) block,
// 64 output pixels x 64 channels, K = 9 taps x 32 ci. The x1 REGION
// feeding all 9 taps (6 rows x 26 cols x 32 ci, channels-last) stages
// once per block — ONE barrier, then 288 MFMAs with the B operand
// (reshaped weights, L2-resident: 74 KB per pair shared by 9*B blocks)
// loaded straight to registers. z2 lands CHANNELS-LAST in zz2.
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_fwd_mfma(CnnArgs a) {
  const int pt = blockIdx.x % 9;           // 576 / 64
  const int gb = blockIdx.x / 9;
  const int g = gb / a.B;
  const int b = gb - g * a.B;
  if (b >= step_n(a, g)) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  // region: x1 rows [r0, r0+6) x 26 cols, channels-last, +1 padded
  __shared__ __attribute__((aligned(16))) float sR[6 * S1][C1 + 1];
  const int p0 = pt * 64;
  const int r0 = p0 / S2;
  const float* x1 = a.x1 + (((long long)g * a.B + b) * 676
                            + (long long)r0 * S1) * C1;
  {
    // last tile: x1 only has rows r0..25 — rows past it are never read
    // by the compute (oy <= 25) but must not be FETCHED (OOB)
    const int nrow = min(6 * S1, (S1 - r0) * S1);
    const int r8 = tid >> 5, kk = tid & 31;
    for (int rr = r8; rr < 6 * S1; rr += 8)
      sR[rr][kk] = (rr < nrow) ? x1[(long long)rr * C1 + kk] : 0.f;
  }
  __syncthreads();
  f32x4 acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};
  // this lane's A row: output pixel p = p0 + wv*16 + li
  const int p = p0 + wv * 16 + li;
  const int arow = (p / S2 - r0) * S1 + (p - (p / S2) * S2);
  const float* wt = a.wtf + (long long)g * 9 * 2048;
  for (int kyx = 0; kyx < 9; ++kyx) {
    const int ky = kyx / 3, kx = kyx - (kyx / 3) * 3;
    const int off = arow + ky * S1 + kx;
    const float* wk = wt + kyx * 2048;
#pragma unroll
    for (int ks = 0; ks < C1 / 4; ++ks) {
      const float av = sR[off][ks * 4 + lk];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
        acc[ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            av, wk[(ks * 4 + lk) * C2 + ct * 16 + li], acc[ct], 0, 0, 0);
    }
  }
  const float* bias = a.work + (long long)g * a.P + OFF_B2C;
  float* z2 = a.zz2 + ((long long)g * a.B + b) * Z2N;
#pragma unroll
  for (int ct = 0; ct < 4; ++ct) {
    const int co = ct * 16 + li;
    const float bb = bias[co];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int pp = p0 + wv * 16 + lk * 4 + r;
      z2[(long long)pp * C2 + co] = acc[ct][r] + bb;
    }
  }
}

// conv2 forward, 128-pixel tiles (the dgrad-v3 trick applied to fwd):
// each wave owns TWO 16-pixel M-fragments, so every weight load feeds
// two MFMAs (fwd was issue-stalled 0.65 with one global per MFMA) and
// each wave runs eight accumulator chains. Region = 8 input rows,
// 27.5 KB LDS.
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_fwd_mfma2(CnnArgs a) {
  const int pt = blockIdx.x % 5;           // ceil(576 / 128)
  const int gb = blockIdx.x / 5;
  const int g = gb / a.B;
  const int b = gb - g * a.B;
  if (b >= step_n(a, g)) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  __shared__ __attribute__((aligned(16))) float sR[8 * S1][C1 + 1];
  const int p0 = pt * 128;
  const int r0 = p0 / S2;
  const float* x1 = a.x1 + (((long long)g * a.B + b) * 676
                            + (long long)r0 * S1) * C1;
  {
    const int nrow = min(8 * S1, (S1 - r0) * S1);
    const int r8 = tid >> 5, kk = tid & 31;
    for (int rr = r8; rr < 8 * S1; rr += 8)
      sR[rr][kk] = (rr < nrow) ? x1[(long long)rr * C1 + kk] : 0.f;
  }
  __syncthreads();
  f32x4 acc[2][4];
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int t = 0; t < 4; ++t) acc[mf][t] = {0.f, 0.f, 0.f, 0.f};
  const int p_a = p0 + wv * 32 + li;
  const int p_b = p_a + 16;
  const int arow_a = (p_a / S2 - r0) * S1 + (p_a - (p_a / S2) * S2);
  const int arow_b = (p_b / S2 - r0) * S1 + (p_b - (p_b / S2) * S2);
  const float* wt = a.wtf + (long long)g * 9 * 2048;
  for (int kyx = 0; kyx < 9; ++kyx) {
    const int ky = kyx / 3, kx = kyx - (kyx / 3) * 3;
    const int offa = arow_a + ky * S1 + kx;
    const int offb = arow_b + ky * S1 + kx;
    const float* wk = wt + kyx * 2048;
#pragma unroll
    for (int ks = 0; ks < C1 / 4; ++ks) {
      const float av0 = sR[offa][ks * 4 + lk];
      const float av1 = sR[offb][ks * 4 + lk];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        const float wv_ = wk[(ks * 4 + lk) * C2 + ct * 16 + li];
        acc[0][ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(av0, wv_,
                                                          acc[0][ct],
                                                          0, 0, 0);
        acc[1][ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(av1, wv_,
                                                          acc[1][ct],
                                                          0, 0, 0);
      }
    }
  }
  const float* bias = a.work + (long long)g * a.P + OFF_B2C;
  float* z2 = a.zz2 + ((long long)g * a.B + b) * Z2N;
#pragma unroll
  for (int ct = 0; ct < 4; ++ct) {
    const int co = ct * 16 + li;
    const float bb = bias[co];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int pp = p0 + wv * 32 + mf * 16 + lk * 4 + r;
        if (pp < Z2N / C2)
          z2[(long long)pp * C2 + co] = acc[mf][ct][r] + bb;
      }
  }
}

// maxpool + dropout1 over the channels-last z2, block per (g, b):
// z2 rows stage through LDS (coalesced channels-last reads), the pooled
// a2/pidx tiles assemble in LDS and store with one coalesced pass in the
// torch-flatten [c][py][px] layout
extern "C" __global__ __launch_bounds__(WG)
void cnn_pool_fwd(CnnArgs a) {
  const int g = blockIdx.x / a.B;
  const int b = blockIdx.x - g * a.B;
  if (b >= step_n(a, g)) return;
  const int tid = threadIdx.x;
  __shared__ __attribute__((aligned(16))) float srow[2][2 * S2 * C2];
  __shared__ __attribute__((aligned(16))) float sa2[NF];
  __shared__ unsigned char spidx[NF];
  const float* z2 = a.zz2 + ((long long)g * a.B + b) * Z2N;
  // double-buffered row-pair staging: the next pooled row's z2 loads
  // issue before this row's max pass, land after it
  float rst[12];
#define PF_LOAD(py)                                                     \
  _Pragma("unroll") for (int jj = 0; jj < 12; ++jj)                      \
    rst[jj] = z2[(long long)(2 * (py)) * S2 * C2 + tid + jj * WG];
#define PF_WRITE(buf)                                                   \
  _Pragma("unroll") for (int jj = 0; jj < 12; ++jj)                      \
    srow[buf][tid + jj * WG] = rst[jj];
  PF_LOAD(0);
  PF_WRITE(0);
  __syncthreads();
  int cur = 0;
  for (int py = 0; py < SP; ++py) {
    if (py + 1 < SP) { PF_LOAD(py + 1); }
    for (int q = tid; q < C2 * SP; q += WG) {
      const int c = q / SP;
      const int px = q - c * SP;
      float best = -1e30f;
      int arg = 0;
#pragma unroll
      for (int dy = 0; dy < 2; ++dy)
#pragma unroll
        for (int dx = 0; dx < 2; ++dx) {
          const float z = srow[cur][(dy * S2 + 2 * px + dx) * C2 + c];
          if (z > best) { best = z; arg = dy * 2 + dx; }
        }
      sa2[c * (SP * SP) + py * SP + px] = best;
      spidx[c * (SP * SP) + py * SP + px] = (unsigned char)arg;
    }
    if (py + 1 < SP) { PF_WRITE(cur ^ 1); }
    __syncthreads();
    cur ^= 1;
  }
#undef PF_LOAD
#undef PF_WRITE
  float* a2o = a.a2 + ((long long)g * a.B + b) * NF;
  unsigned char* po = a.pidx + ((long long)g * a.B + b) * NF;
  for (int e = tid; e < NF; e += WG) {
    a2o[e] = sa2[e] * drop_scale(a, 0, g, b, e);
    po[e] = spidx[e];
  }
}

// fc1 forward as MFMA GEMM with K split over blocks (fills the chip at
// small fleets): block (g, mtile, ks) computes partial z1 for 64 batch
// rows x 128 h over K-range [ks*NF/KS, ...). Partials land in z1part.
#define FC1_KS 8

extern "C" __global__ __launch_bounds__(WG)
void cnn_fc1_fwd_mfma(CnnArgs a) {
  const int mtiles = (a.B + 63) / 64;
  const int ks = blockIdx.x % FC1_KS;
  const int rest = blockIdx.x / FC1_KS;
  const int mt = rest % mtiles;
  const int g = rest / mtiles;
  const int n = step_n(a, g);
  if (mt * 64 >= n) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  __shared__ __attribute__((aligned(16))) float sA[2][64][EVAL_BK + 1];
  __shared__ __attribute__((aligned(16))) float sB[2][EVAL_BK][NH + 1];
  f32x4 acc[8];
#pragma unroll
  for (int t = 0; t < 8; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};
  const float* wp = a.work + (long long)g * a.P + OFF_W1F;
  const float* a2 = a.a2 + ((long long)g * a.B + (long long)mt * 64) * NF;
  const int mlen = min(64, n - mt * 64);
  const int r8 = tid >> 5, kk = tid & 31;
  const int k_lo = ks * (NF / FC1_KS), k_hi = (ks + 1) * (NF / FC1_KS);
  float ra[8], rb[16];
#define FC_LOAD(k0)                                                     \
  _Pragma("unroll") for (int j = 0; j < 8; ++j) {                        \
    const int rr = r8 + j * 8;                                           \
    ra[j] = (rr < mlen) ? a2[(long long)rr * NF + (k0) + kk] : 0.f;      \
  }                                                                      \
  _Pragma("unroll") for (int j = 0; j < 16; ++j)                         \
    rb[j] = wp[(long long)(r8 + j * 8) * NF + (k0) + kk];
#define FC_WRITE(buf)                                                   \
  _Pragma("unroll") for (int j = 0; j < 8; ++j)                          \
    sA[buf][r8 + j * 8][kk] = ra[j];                                     \
  _Pragma("unroll") for (int j = 0; j < 16; ++j)                         \
    sB[buf][kk][r8 + j * 8] = rb[j];
  FC_LOAD(k_lo);
  FC_WRITE(0);
  __syncthreads();
  int cur = 0;
  for (int k0 = k_lo; k0 < k_hi; k0 += EVAL_BK) {
    if (k0 + EVAL_BK < k_hi) { FC_LOAD(k0 + EVAL_BK); }
#pragma unroll
    for (int kq = 0; kq < EVAL_BK / 4; ++kq) {
      const float av = sA[cur][wv * 16 + li][kq * 4 + lk];
#pragma unroll
      for (int ct = 0; ct < 8; ++ct)
        acc[ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            av, sB[cur][kq * 4 + lk][ct * 16 + li], acc[ct], 0, 0, 0);
    }
    if (k0 + EVAL_BK < k_hi) { FC_WRITE(cur ^ 1); }
    __syncthreads();
    cur ^= 1;
  }
#undef FC_LOAD
#undef FC_WRITE
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mrow = wv * 16 + lk * 4 + r;
      if (mrow < mlen) {
        const long long b = (long long)mt * 64 + mrow;
        a.z1part[(((long long)g * FC1_KS + ks) * a.B + b) * NH
                 + ct * 16 + li] = acc[ct][r];
      }
    }
  }
}

// combine fc1 K-split partials + bias, apply relu + dropout2
extern "C" __global__ __launch_bounds__(WG)
void cnn_fc1_act(CnnArgs a) {
  const long long total = (long long)a.G * a.B * NH;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)a.B * NH));
    const long long r = q - (long long)g * a.B * NH;
    const int b = (int)(r / NH);
    if (b >= step_n(a, g)) continue;
    const int h = (int)(r - (long long)b * NH);
    float z = a.work[(long long)g * a.P + OFF_B1F + h];
#pragma unroll
    for (int ks = 0; ks < FC1_KS; ++ks)
      z += a.z1part[(((long long)g * FC1_KS + ks) * a.B + b) * NH + h];
    a.z1[q] = z;
    const float rl = z > 0.f ? z : 0.f;
    a.a1[q] = rl * drop_scale(a, 1, g, b, h);
  }
}

// fc2 + softmax (in-graph model output s) + CE-on-s gradient -> dz2
// dL/ds = (softmax(s) - onehot(y)) / n; dL/dz2 = s*(dL/ds - sum(dL/ds*s))
//
// ONE WAVE per sample: lane l accumulates logit l (O <= 64 always), the
// a1 row loads once into two coalesced registers per lane, and both
// softmaxes run as 6-step shuffle reductions — no per-thread O-arrays
// (the previous thread-per-sample form kept z2/s/ds[64] in scratch:
// 133 us/dispatch of spill traffic at the config-3 probe shape).
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int m = 1; m < 64; m <<= 1) v += __shfl_xor(v, m, 64);
  return v;
}
__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int m = 1; m < 64; m <<= 1) v = fmaxf(v, __shfl_xor(v, m, 64));
  return v;
}

extern "C" __global__ __launch_bounds__(WG)
void cnn_head_fwd(CnnArgs a) {
  const long long total = (long long)a.G * a.B;
  const int l = threadIdx.x & 63;
  for (long long q = (long long)blockIdx.x * (WG / 64) + (threadIdx.x >> 6);
       q < total; q += (long long)gridDim.x * (WG / 64)) {
    const int g = (int)(q / a.B);
    const int b = (int)(q - (long long)g * a.B);
    const int n = step_n(a, g);
    if (b >= n) continue;
    const float inv_n = 1.f / (float)n;
    const float* w = a.work + (long long)g * a.P;
    const float* a1 = a.a1 + q * NH;
    const int yi = (int)a.y[step_o(a, g) + b];
    const float r0 = a1[l], r1 = a1[64 + l];
    // lane o collects logit o: each o-round is two coalesced weight
    // loads + two fmas + a wave reduce
    float zlane = 0.f;
    for (int o = 0; o < a.O; ++o) {
      const float* wo = w + OFF_W2F + (long long)o * NH;
      const float v = wave_sum(fmaf(r0, wo[l], r1 * wo[64 + l]));
      if (l == o) zlane = v;
    }
    const bool live = l < a.O;
    float z = live ? zlane + w[OFF_W2F + (long long)a.O * NH + l] : -1e30f;
    const float zmax = wave_max(z);
    float e = live ? __expf(z - zmax) : 0.f;
    const float s = e / wave_sum(e);               // model output s[l]
    // CE(log_softmax(s), y): q2 = softmax(s)
    const float smax = wave_max(live ? s : -1e30f);
    float e2 = live ? __expf(s - smax) : 0.f;
    float q2 = e2 / wave_sum(e2);
    if (l == yi) q2 -= 1.f;
    const float ds = q2 * inv_n;
    const float dot = wave_sum(live ? ds * s : 0.f);
    if (live) a.dz2[q * a.O + l] = s * (ds - dot);
  }
}

// ---------------------------------------------------------------------------
// backward
// ---------------------------------------------------------------------------

// fc2 wgrad + bias grad: thread per (g, o, h) (+ o for bias when h == 0)
extern "C" __global__ __launch_bounds__(WG)
void cnn_fc2_wgrad(CnnArgs a) {
  const long long total = (long long)a.G * a.O * NH;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)a.O * NH));
    const long long r = q - (long long)g * a.O * NH;
    const int o = (int)(r / NH);
    const int h = (int)(r - (long long)o * NH);
    const int n = step_n(a, g);
    if (n == 0) continue;
    float acc = 0.f, accb = 0.f;
    for (int b = 0; b < n; ++b) {
      const float d = a.dz2[((long long)g * a.B + b) * a.O + o];
      acc = fmaf(d, a.a1[((long long)g * a.B + b) * NH + h], acc);
      if (h == 0) accb += d;
    }
    float* gr = a.grad + (long long)g * a.P;
    gr[OFF_W2F + (long long)o * NH + h] = acc;
    if (h == 0) gr[OFF_W2F + (long long)a.O * NH + o] = accb;
  }
}

// fc2 dgrad -> through dropout2 + relu -> dz1: thread per (g, b, h)
extern "C" __global__ __launch_bounds__(WG)
void cnn_fc2_dgrad(CnnArgs a) {
  const long long total = (long long)a.G * a.B * NH;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)a.B * NH));
    const long long r = q - (long long)g * a.B * NH;
    const int b = (int)(r / NH);
    if (b >= step_n(a, g)) continue;
    const int h = (int)(r - (long long)b * NH);
    const float* w = a.work + (long long)g * a.P + OFF_W2F;
    float acc = 0.f;
    for (int o = 0; o < a.O; ++o)
      acc = fmaf(a.dz2[((long long)g * a.B + b) * a.O + o],
                 w[(long long)o * NH + h], acc);
    const float zv = a.z1[q];
    a.dz1[q] = zv > 0.f ? acc * drop_scale(a, 1, g, b, h) : 0.f;
  }
}

// fc1 wgrad as MFMA GEMM: dW1f[h][j] = sum_b dz1[b][h] a2[b][j] —
// block per (g, jtile of 128 columns), 64 (h-tile, j-tile) outputs
// split over 4 waves, K = batch staged in 16-sample chunks (both
// operands come in [b][.] layout, so the A fragment reads the staged
// dz1 tile transposed in place). Exclusive-owner writes.
extern "C" __global__ __launch_bounds__(WG)
void cnn_fc1_wgrad(CnnArgs a) {
  const int jt = blockIdx.x % (NF / 128);
  const int g = blockIdx.x / (NF / 128);
  const int n = step_n(a, g);
  if (n == 0) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  __shared__ __attribute__((aligned(16))) float sDZ[16][NH + 1];
  __shared__ __attribute__((aligned(16))) float sA2[16][129];
  f32x4 acc[16];
#pragma unroll
  for (int t = 0; t < 16; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};
  const float* dz1 = a.dz1 + (long long)g * a.B * NH;
  const float* a2 = a.a2 + (long long)g * a.B * NF + jt * 128;
  const int j = tid & 127;
  const int b2 = tid >> 7;                // 2 stager rows per pass
  for (int b0 = 0; b0 < n; b0 += 16) {
    for (int bb = b2; bb < 16; bb += 2) {
      const bool ok = b0 + bb < n;
      sDZ[bb][j] = ok ? dz1[(long long)(b0 + bb) * NH + j] : 0.f;
      sA2[bb][j] = ok ? a2[(long long)(b0 + bb) * NF + j] : 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
#pragma unroll
      for (int t = 0; t < 16; ++t) {
        const int tile = wv * 16 + t;
        const int rt = tile >> 3, ct = tile & 7;
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            sDZ[ks * 4 + lk][rt * 16 + li],
            sA2[ks * 4 + lk][ct * 16 + li], acc[t], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  float* gr = a.grad + (long long)g * a.P + OFF_W1F;
#pragma unroll
  for (int t = 0; t < 16; ++t) {
    const int tile = wv * 16 + t;
    const int rt = tile >> 3, ct = tile & 7;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int h = rt * 16 + lk * 4 + r;
      gr[(long long)h * NF + jt * 128 + ct * 16 + li] = acc[t][r];
    }
  }
}

// fc1 bias grad: thread per (g, h)
extern "C" __global__ __launch_bounds__(WG)
void cnn_fc1_bias_grad(CnnArgs a) {
  const long long total = (long long)a.G * NH;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / NH);
    const int h = (int)(q - (long long)g * NH);
    const int n = step_n(a, g);
    if (n == 0) continue;
    float s = 0.f;
    for (int b = 0; b < n; ++b)
      s += a.dz1[((long long)g * a.B + b) * NH + h];
    a.grad[(long long)g * a.P + OFF_B1F + h] = s;
  }
}

// fc1 dgrad as MFMA GEMM: da2[b][j] = sum_h dz1[b][h] W1f[h][j] —
// block per (g, b-tile of 64, jtile of 128), K = 128 in 32-deep chunks
extern "C" __global__ __launch_bounds__(WG)
void cnn_fc1_dgrad(CnnArgs a) {
  const int jt = blockIdx.x % (NF / 128);
  const int rest = blockIdx.x / (NF / 128);
  const int mtiles = (a.B + 63) / 64;
  const int mt = rest % mtiles;
  const int g = rest / mtiles;
  const int n = step_n(a, g);
  if (mt * 64 >= n) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  __shared__ __attribute__((aligned(16))) float sA[64][33];
  __shared__ __attribute__((aligned(16))) float sB[32][129];
  f32x4 acc[8];
#pragma unroll
  for (int t = 0; t < 8; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};
  const int mlen = min(64, n - mt * 64);
  const float* dz1 = a.dz1 + ((long long)g * a.B + mt * 64) * NH;
  const float* w = a.work + (long long)g * a.P + OFF_W1F + jt * 128;
  const int r8 = tid >> 5, kk = tid & 31;
  const int j = tid & 127, h2 = tid >> 7;
  for (int k0 = 0; k0 < NH; k0 += 32) {
    for (int rr = r8; rr < 64; rr += 8)
      sA[rr][kk] = (rr < mlen)
          ? dz1[(long long)rr * NH + k0 + kk] : 0.f;
    for (int hh = h2; hh < 32; hh += 2)
      sB[hh][j] = w[(long long)(k0 + hh) * NF + j];
    __syncthreads();
#pragma unroll
    for (int ks = 0; ks < 8; ++ks) {
      const float av = sA[wv * 16 + li][ks * 4 + lk];
#pragma unroll
      for (int ct = 0; ct < 8; ++ct)
        acc[ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            av, sB[ks * 4 + lk][ct * 16 + li], acc[ct], 0, 0, 0);
    }
    __syncthreads();
  }
  float* da2 = a.da2 + ((long long)g * a.B + mt * 64) * NF + jt * 128;
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wv * 16 + lk * 4 + r;
      if (row < mlen)
        da2[(long long)row * NF + ct * 16 + li] = acc[ct][r];
    }
  }
}

// pool backward, block per (g, b): da2 reads coalesced (flatten layout,
// dropout reapplied), routed dz2 writes coalesced channels-last via an
// LDS-staged transpose; the conv2 bias grad (sum of routed dz) emits a
// per-sample partial (b2part) that the wgrad reduce sums deterministically
extern "C" __global__ __launch_bounds__(WG)
void cnn_pool_bwd(CnnArgs a) {
  const int g = blockIdx.x / a.B;
  const int b = blockIdx.x - g * a.B;
  if (b >= step_n(a, g)) return;
  const int tid = threadIdx.x;
  __shared__ __attribute__((aligned(16))) float sda[NF];
  __shared__ float sbias[C2];
  const float* da2 = a.da2 + ((long long)g * a.B + b) * NF;
  for (int e = tid; e < NF; e += WG)
    sda[e] = da2[e] * drop_scale(a, 0, g, b, e);
  if (tid < C2) sbias[tid] = 0.f;
  __syncthreads();
  const unsigned char* pidx = a.pidx + ((long long)g * a.B + b) * NF;
  float* dzc = a.zz2 + ((long long)g * a.B + b) * Z2N;
  float bacc = 0.f;
  const int c = tid & 63;            // fixed per thread (WG % 64 == 0)
  for (int q = tid; q < Z2N; q += WG) {
    const int p2 = q >> 6;           // z2 pixel
    const int oy = p2 / S2, ox = p2 - (p2 / S2) * S2;
    const int e = c * (SP * SP) + (oy >> 1) * SP + (ox >> 1);
    const int arg = (oy & 1) * 2 + (ox & 1);
    const float v = ((int)pidx[e] == arg) ? sda[e] : 0.f;
    dzc[q] = v;
    bacc += v;
  }
  // per-channel reduce (4 threads share each c), then a per-sample
  // partial into the dz2 scratch ([G, B, 64], free after fc2 backward);
  // cnn_conv2_wgrad_reduce sums it deterministically
  for (int off = 192; off > 0; off -= 64) {
    __syncthreads();
    if (tid >= off && tid < off + 64) sbias[c] += bacc;
  }
  __syncthreads();
  if (tid < C2)
    a.b2part[((long long)g * a.B + b) * C2 + tid] = sbias[tid] + bacc;
}

// conv2 wgrad as MFMA GEMM: dW[(kyx, ci)][co] = sum_m A[m, ci] dz[m, co]
// with m = (b, pixel). Block per (g, ms): for each owned 64-pixel
// m-tile, the x1 REGION (feeds all 9 taps) and the dz tile stage ONCE;
// each wave accumulates 18 of the 72 (kyx, ci-tile, co-tile) outputs
// (acc stays in registers across the whole m-range). Partials land in
// w2part[g][ms] and reduce deterministically. The m-split count is a
// runtime knob (a.w2ms) so small fleets still fill 256 CUs.
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_wgrad_mfma(CnnArgs a) {
  const int ms = blockIdx.x % a.w2ms;
  const int g = blockIdx.x / a.w2ms;
  const int n = step_n(a, g);
  float* part = a.w2part + ((long long)g * a.w2ms + ms) * (9 * 2048);
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  // all staging by async DMA (global_load_lds): both operands are read
  // stride-1 per lane group, so LINEAR LDS images need no padding or
  // swizzle. Region [160 rows x 32 ci] (rows 156..159 are never read),
  // dz tile [64 m x 64 co]; both double-buffered with counted waits +
  // raw barriers so the next m-tile streams under this tile's MFMAs.
  // ONE shared object (a second __shared__ makes hipcc drain vmcnt(0)
  // before every ds_read beside an in-flight glds — guide §5 trap (a))
  __shared__ __attribute__((aligned(16))) float ldsw[2 * 160 * C1
                                                     + 2 * 64 * C2];
  float (*sR)[160 * C1] = (float (*)[160 * C1])ldsw;
  float (*sD)[64 * C2] = (float (*)[64 * C2])(ldsw + 2 * 160 * C1);
  f32x4 acc[18];
#pragma unroll
  for (int t = 0; t < 18; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};
  const long long mtiles = ((long long)n * (S2 * S2) + 63) / 64;
  // per wave per tile: 5 region glds + 4 dz glds (1 KiB each)
#define WG_ISSUE(MT, BUF)                                                \
  {                                                                      \
    const long long m0_ = (MT) * 64;                                     \
    const int b_ = (int)(m0_ / (S2 * S2));                               \
    const int p0_ = (int)(m0_ - (long long)b_ * (S2 * S2));              \
    const int r0_ = p0_ / S2;                                            \
    const int nrow_ = (S1 - r0_) * S1;                                   \
    const long long gb_ = (long long)g * a.B + b_;                       \
    const float* x1_ = a.x1 + (gb_ * 676 + (long long)r0_ * S1) * C1;    \
    const float* dz_ = a.zz2 + (gb_ * 576 + p0_) * C2;                   \
    _Pragma("unroll") for (int gq = 0; gq < 5; ++gq) {                   \
      const int f0_ = (wv * 5 + gq) * 256;      /* region floats */      \
      int srow_ = (f0_ + l * 4) / C1;           /* source arow */        \
      srow_ = srow_ < nrow_ ? srow_ : nrow_ - 1;                         \
      const int sch_ = (f0_ + l * 4) & 31;                               \
      __builtin_amdgcn_global_load_lds(                                  \
          (const __attribute__((address_space(1))) unsigned int*)        \
              (x1_ + (long long)srow_ * C1 + sch_),                      \
          (__attribute__((address_space(3))) unsigned int*)              \
              (&sR[BUF][f0_]),                                           \
          16, 0, 0);                                                     \
    }                                                                    \
    _Pragma("unroll") for (int gq = 0; gq < 4; ++gq) {                   \
      const int f0_ = (wv * 4 + gq) * 256;                               \
      __builtin_amdgcn_global_load_lds(                                  \
          (const __attribute__((address_space(1))) unsigned int*)        \
              (dz_ + f0_ + l * 4),                                       \
          (__attribute__((address_space(3))) unsigned int*)              \
              (&sD[BUF][f0_]),                                           \
          16, 0, 0);                                                     \
    }                                                                    \
  }
  int cur = 0;
  if (ms < mtiles) { WG_ISSUE(ms, 0); }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  for (long long mt = ms; mt < mtiles; mt += a.w2ms) {
    const int p0 = (int)((mt * 64) % (S2 * S2));
    const int r0 = p0 / S2;
    if (mt + a.w2ms < mtiles) { WG_ISSUE(mt + a.w2ms, cur ^ 1); }
    int parow[16];
#pragma unroll
    for (int km = 0; km < 16; ++km) {
      const int q = p0 + km * 4 + lk;
      parow[km] = (q / S2 - r0) * S1 + (q - (q / S2) * S2);
    }
    const float* bR = sR[cur];
    const float* bD = sD[cur];
#pragma unroll
    for (int kyx = 0; kyx < 9; ++kyx) {
      const int kyoff = (kyx / 3) * S1 + (kyx - (kyx / 3) * 3);
#pragma unroll
      for (int tt = 0; tt < 8; ++tt) {
        if ((kyx * 8 + tt) % 4 != wv) continue;
        const int slot = (kyx * 8 + tt) / 4;
        const int rt = tt >> 2, ct = tt & 3;
#pragma unroll
        for (int km = 0; km < 16; ++km) {
          acc[slot] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              bR[(parow[km] + kyoff) * C1 + rt * 16 + li],
              bD[(km * 4 + lk) * C2 + ct * 16 + li], acc[slot],
              0, 0, 0);
        }
      }
    }
    if (mt + a.w2ms < mtiles) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }
#undef WG_ISSUE
#pragma unroll
  for (int slot = 0; slot < 18; ++slot) {
    const int kyx = (slot * 4 + wv) / 8;
    const int tt = (slot * 4 + wv) % 8;
    const int rt = tt >> 2, ct = tt & 3;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int ci = rt * 16 + lk * 4 + r;
      part[kyx * 2048 + ci * C2 + ct * 16 + li] = acc[slot][r];
    }
  }
}

// reduce the conv2 wgrad msplit partials into the (co, ci, ky, kx) grad;
// the tail entries (one per (g, co)) reduce the bias partials from
// pool_bwd over the batch — all deterministic
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_wgrad_reduce(CnnArgs a) {
  const long long per_g = 9 * 2048 + C2;
  const long long total = (long long)a.G * per_g;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / per_g);
    const int r = (int)(q - (long long)g * per_g);
    const int n = step_n(a, g);
    if (n == 0) continue;
    if (r >= 9 * 2048) {
      const int co = r - 9 * 2048;
      float s = 0.f;
      for (int b = 0; b < n; ++b)
        s += a.b2part[((long long)g * a.B + b) * C2 + co];
      a.grad[(long long)g * a.P + OFF_B2C + co] = s;
      continue;
    }
    const int kyx = r / 2048;
    const int t = r - kyx * 2048;
    const int ci = t / C2, co = t - (t / C2) * C2;
    const float* part = a.w2part + (long long)g * a.w2ms * 9 * 2048
                        + kyx * 2048 + t;
    float s = 0.f;
    for (int ms = 0; ms < a.w2ms; ++ms)
      s += part[(long long)ms * 9 * 2048];
    a.grad[(long long)g * a.P + OFF_W2C + (co * C1 + ci) * 9 + kyx] = s;
  }
}

// conv2 dgrad as MFMA GEMM: dx1[m=(b,y,x)][ci] over K = (kyx, co).
// ONE block per (g, b) walks all 11 64-pixel tiles with EVERYTHING
// LDS-resident via global_load_lds (async DMA): the full 74 KB weight
// tensor stages once in the prologue, and the 37 KB dz region
// double-buffers so the next tile streams while the current tile's
// 288 MFMAs issue (counted-wait + raw barriers; __syncthreads or any
// ordinary global load in the loop would drain the DMA queue — guide
// T3/T4 and the §5 mixing-load-kinds trap). The region image is linear
// (glds requirement): bank spread comes from a 4-float channel
// rotation keyed on the row-col index, applied identically at the DMA
// source address and at read time; out-of-bounds taps resolve by
// operand predication instead of staged zero borders.
#define DG_RC (6 * S2)              // 144 row-cols per tile region

extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_dgrad_mfma(CnnArgs a) {
  const int gb = blockIdx.x;
  const int g = gb / a.B;
  const int b = gb - g * a.B;
  if (b >= step_n(a, g)) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  // carve: sB [9*2048] then sD [2][DG_RC][64]
  extern __shared__ __attribute__((aligned(16))) float lds_[];
  float* sB = lds_;
  float* sD = lds_ + 9 * 2048;
  const float* dz = a.zz2 + ((long long)g * a.B + b) * Z2N;
  const float* wt = a.wtd + (long long)g * 9 * 2048;
  float* dx1 = a.dx1 + ((long long)g * a.B + b) * X1N;

  // prologue: the whole wtd[9][64][32] tensor -> LDS (18 glds per wave)
  {
    const int base = wv * (18 * 256);       // floats
#pragma unroll
    for (int gq = 0; gq < 18; ++gq)
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)
              (wt + base + gq * 256 + l * 4),
          (__attribute__((address_space(3))) unsigned int*)
              (sB + base + gq * 256),
          16, 0, 0);
  }

  // each wave DMAs rows {wv, wv + 4 partial} of the 6-row region: the
  // 144 row-cols split as 36 glds (4 arows each) -> 9 per wave; invalid
  // rows load from a clamped source row and are masked at use
#define DG_ISSUE(mt, buf)                                                \
  {                                                                      \
    const int y0_ = (mt) * 64 / S1;                                      \
    _Pragma("unroll") for (int gq = 0; gq < 9; ++gq) {                   \
      const int arow0_ = (wv * 9 + gq) * 4;                              \
      const int arow_l = arow0_ + (l >> 4);                              \
      const int row_ = arow_l / S2;                                      \
      const int col_ = arow_l - row_ * S2;                               \
      int dzrow_ = y0_ - 2 + row_;  /* region = dz rows [y0-2, y0+4) */  \
      dzrow_ = dzrow_ < 0 ? 0 : (dzrow_ > S2 - 1 ? S2 - 1 : dzrow_);     \
      const int chp_ = (l & 15) * 4;                                     \
      const int chs_ = (chp_ - 4 * arow_l) & 63;                         \
      __builtin_amdgcn_global_load_lds(                                  \
          (const __attribute__((address_space(1))) unsigned int*)        \
              (dz + ((long long)dzrow_ * S2 + col_) * C2 + chs_),        \
          (__attribute__((address_space(3))) unsigned int*)              \
              (sD + (buf) * (DG_RC * 64) + arow0_ * 64),                 \
          16, 0, 0);                                                     \
    }                                                                    \
  }

  DG_ISSUE(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  int cur = 0;
  for (int mt = 0; mt < 11; ++mt) {
    if (mt + 1 < 11) { DG_ISSUE(mt + 1, cur ^ 1); }
    const int p0 = mt * 64;
    const int y0 = p0 / S1;
    const int p = p0 + wv * 16 + li;        // this lane's x1 pixel
    const int y = p / S1, x = p - (p / S1) * S1;
    f32x4 acc[2];
    acc[0] = f32x4{0.f, 0.f, 0.f, 0.f};
    acc[1] = f32x4{0.f, 0.f, 0.f, 0.f};
    const float* base = sD + cur * (DG_RC * 64);
    for (int kyx = 0; kyx < 9; ++kyx) {
      const int ky = kyx / 3, kx = kyx - (kyx / 3) * 3;
      const float* wk = sB + kyx * 2048;
      const int oy = y - ky, ox = x - kx;
      // region = dz rows [y0-2, y0+4): every VALID tap of this tile's
      // pixels lands inside (oy in [y-2, y], y in [y0, y0+2])
      const bool ok = oy >= 0 && oy < S2 && ox >= 0 && ox < S2;
      const int arow = ok ? (oy - y0 + 2) * S2 + ox : 0;
      const int rot = 4 * arow;
      const float* rowp = base + arow * 64;
#pragma unroll
      for (int ks = 0; ks < C2 / 4; ++ks) {
        const int co = ks * 4 + lk;
        float av = rowp[(co + rot) & 63];
        av = ok ? av : 0.f;
#pragma unroll
        for (int ct = 0; ct < 2; ++ct)
          acc[ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              av, wk[co * C1 + ct * 16 + li], acc[ct], 0, 0, 0);
      }
    }
#pragma unroll
    for (int ct = 0; ct < 2; ++ct)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int pix = p0 + wv * 16 + lk * 4 + r;
        if (pix < 676)
          dx1[(long long)pix * C1 + ct * 16 + li] = acc[ct][r];
      }
    if (mt + 1 < 11) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }
#undef DG_ISSUE
}

// conv2 dgrad v2 — the fwd-shaped decomposition: ONE block per
// (g, b, mtile of 11), mirroring cnn_conv2_fwd_mfma. The block's 6-row
// dz region (37 KB, zero-bordered, +1-padded) stages once with ordinary
// loads and ONE barrier; the B operand (wtd, 74 KB per pair) reads
// straight from L2 inside the MFMA loop exactly like fwd's wtf. v1
// keeps everything (weights + region) LDS-resident via the glds DMA
// pipeline, which is elegant but needs 144 KB of LDS -> ONE block
// (4 waves) per CU, and its inter-tile barrier + vmcnt waits run
// unhidden (PMC: 57% parked, profiles/pmc5_summary.json). This shape
// fits 4 blocks/CU, so parked time overlaps across blocks.
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_dgrad_mfma2(CnnArgs a) {
  const int mt = blockIdx.x % 11;
  const int gb = blockIdx.x / 11;
  const int g = gb / a.B;
  const int b = gb - g * a.B;
  if (b >= step_n(a, g)) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  // row stride C2+4 = 68 === 4 (mod 32): the MFMA-loop read
  // sD[arow][ks*4+lk] with arow consecutive in li then hits bank
  // 4*li + lk -- an exact 2-way spread (the 64-lane minimum)
  __shared__ __attribute__((aligned(16))) float sD[6 * S2][C2 + 4];
  const float* dz = a.zz2 + ((long long)g * a.B + b) * Z2N;
  const float* wt = a.wtd + (long long)g * 9 * 2048;
  float* dx1 = a.dx1 + ((long long)g * a.B + b) * X1N;
  const int p0 = mt * 64;
  const int y0 = p0 / S1;          // first dx1 pixel row of this tile
  // region = dz rows [y0-2, y0+4): every valid tap of pixels y in
  // [y0, y0+2] lands inside; rows outside [0, S2) stage as zeros
  for (int e = tid; e < 6 * S2 * C2; e += WG) {
    const int rr = e / (S2 * C2);
    const int rem = e - rr * S2 * C2;
    const int cc = rem / C2;
    const int ch = rem - cc * C2;
    const int dzrow = y0 - 2 + rr;
    sD[rr * S2 + cc][ch] =
        (dzrow >= 0 && dzrow < S2)
            ? dz[((long long)dzrow * S2 + cc) * C2 + ch] : 0.f;
  }
  __syncthreads();
  const int p = p0 + wv * 16 + li;        // this lane's dx1 pixel
  const int y = p / S1, x = p - (p / S1) * S1;
  f32x4 acc[2];
  acc[0] = f32x4{0.f, 0.f, 0.f, 0.f};
  acc[1] = f32x4{0.f, 0.f, 0.f, 0.f};
  for (int kyx = 0; kyx < 9; ++kyx) {
    const int ky = kyx / 3, kx = kyx - (kyx / 3) * 3;
    const int oy = y - ky, ox = x - kx;
    // x borders (S1=26 > S2=24) still need operand predication; the
    // staged y borders are already zero
    const bool ok = oy >= 0 && oy < S2 && ox >= 0 && ox < S2;
    const int arow = ok ? (oy - y0 + 2) * S2 + ox : 0;
    const float* wk = wt + kyx * 2048;
#pragma unroll
    for (int ks = 0; ks < C2 / 4; ++ks) {
      const int co = ks * 4 + lk;
      float av = sD[arow][co];
      av = ok ? av : 0.f;
#pragma unroll
      for (int ct = 0; ct < 2; ++ct)
        acc[ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            av, wk[co * C1 + ct * 16 + li], acc[ct], 0, 0, 0);
    }
  }
#pragma unroll
  for (int ct = 0; ct < 2; ++ct)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int pix = p0 + wv * 16 + lk * 4 + r;
      if (pix < 676)
        dx1[(long long)pix * C1 + ct * 16 + li] = acc[ct][r];
    }
}

// conv2 dgrad v3 — v2 with 128-pixel tiles: each wave owns TWO
// 16-pixel M-fragments, so every weight load feeds two MFMAs (the B
// operand is the per-MFMA global-issue cost in v2) and each wave runs
// four independent accumulator chains. Region = 8 dz rows (52 KB,
// 3 blocks/CU).
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_dgrad_mfma3(CnnArgs a) {
  const int mt = blockIdx.x % 6;          // 676 / 128 -> 6 tiles
  const int gb = blockIdx.x / 6;
  const int g = gb / a.B;
  const int b = gb - g * a.B;
  if (b >= step_n(a, g)) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  __shared__ __attribute__((aligned(16))) float sD[8 * S2][C2 + 4];
  const float* dz = a.zz2 + ((long long)g * a.B + b) * Z2N;
  const float* wt = a.wtd + (long long)g * 9 * 2048;
  float* dx1 = a.dx1 + ((long long)g * a.B + b) * X1N;
  const int p0 = mt * 128;
  const int y0 = p0 / S1;
  // region = dz rows [y0-2, y0+6): pixels y in [y0, y0+5], taps reach
  // oy in [y0-2, y0+5]; rows outside [0, S2) stage as zeros
  for (int e = tid; e < 8 * S2 * C2; e += WG) {
    const int rr = e / (S2 * C2);
    const int rem = e - rr * S2 * C2;
    const int cc = rem / C2;
    const int ch = rem - cc * C2;
    const int dzrow = y0 - 2 + rr;
    sD[rr * S2 + cc][ch] =
        (dzrow >= 0 && dzrow < S2)
            ? dz[((long long)dzrow * S2 + cc) * C2 + ch] : 0.f;
  }
  __syncthreads();
  const int p_a = p0 + wv * 32 + li;       // M-fragment 0
  const int p_b = p_a + 16;                // M-fragment 1
  const int ya = p_a / S1, xa = p_a - ya * S1;
  const int yb = p_b / S1, xb = p_b - yb * S1;
  f32x4 acc[2][2];
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int ct = 0; ct < 2; ++ct) acc[mf][ct] = f32x4{0.f, 0.f, 0.f, 0.f};
  for (int kyx = 0; kyx < 9; ++kyx) {
    const int ky = kyx / 3, kx = kyx - (kyx / 3) * 3;
    const int oya = ya - ky, oxa = xa - kx;
    const int oyb = yb - ky, oxb = xb - kx;
    const bool oka = oya >= 0 && oya < S2 && oxa >= 0 && oxa < S2;
    const bool okb = oyb >= 0 && oyb < S2 && oxb >= 0 && oxb < S2;
    const int ra = oka ? (oya - y0 + 2) * S2 + oxa : 0;
    const int rb = okb ? (oyb - y0 + 2) * S2 + oxb : 0;
    const float* wk = wt + kyx * 2048;
#pragma unroll
    for (int ks = 0; ks < C2 / 4; ++ks) {
      const int co = ks * 4 + lk;
      float av0 = sD[ra][co];
      float av1 = sD[rb][co];
      av0 = oka ? av0 : 0.f;
      av1 = okb ? av1 : 0.f;
      const float wk0 = wk[co * C1 + li];
      const float wk1 = wk[co * C1 + 16 + li];
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(av0, wk0,
                                                       acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(av0, wk1,
                                                       acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(av1, wk0,
                                                       acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(av1, wk1,
                                                       acc[1][1], 0, 0, 0);
    }
  }
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int ct = 0; ct < 2; ++ct)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int pix = p0 + wv * 32 + mf * 16 + lk * 4 + r;
        if (pix < 676)
          dx1[(long long)pix * C1 + ct * 16 + li] = acc[mf][ct][r];
      }
}

// conv1 wgrad, stage 1: per-(g, b) partials into the dz2 scratch region
// (reused: dz2 is [G, B, 64] and conv1 has 288+32=320 grad entries, so
// partials use their own ws buffer c1part [G, B, 320]). Deterministic
// two-stage reduce (no atomics).
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv1_wgrad_part(CnnArgs a) {
  const int g = blockIdx.x / a.B;
  const int b = blockIdx.x - g * a.B;
  if (b >= step_n(a, g)) return;
  const int tid = threadIdx.x;
  __shared__ __attribute__((aligned(16))) float xin[D_IN];
  const float* xs = a.x + (step_o(a, g) + b) * D_IN;
  const float* xm = a.x_mask ? a.x_mask + (long long)g * D_IN : nullptr;
  for (int d = tid; d < D_IN; d += WG)
    xin[d] = xm ? xs[d] * xm[d] : xs[d];
  __syncthreads();
  const float* dx = a.dx1 + ((long long)g * a.B + b) * X1N;  // chans-last
  float* out = a.c1part + ((long long)g * a.B + b) * 320;
  for (int tap = tid; tap < 320; tap += WG) {
    float s = 0.f;
    if (tap < 288) {
      const int c = tap / 9;
      const int k = tap - c * 9;
      const int ky = k / 3, kx = k - (k / 3) * 3;
      for (int y = 0; y < S1; ++y) {
        const float* xr = xin + (y + ky) * IN_W + kx;
        const float* dr = dx + (long long)y * S1 * C1 + c;
        for (int x = 0; x < S1; ++x)
          s = fmaf(xr[x], dr[(long long)x * C1], s);
      }
    } else {
      const int c = tap - 288;
      for (int p = 0; p < S1 * S1; ++p) s += dx[(long long)p * C1 + c];
    }
    out[tap] = s;
  }
}

// conv1 wgrad, stage 2: reduce partials over the batch
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv1_wgrad_reduce(CnnArgs a) {
  const long long total = (long long)a.G * 320;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / 320);
    const int tap = (int)(q - (long long)g * 320);
    const int n = step_n(a, g);
    if (n == 0) continue;
    const float* part = a.c1part + (long long)g * a.B * 320 + tap;
    float s = 0.f;
    for (int b = 0; b < n; ++b) s += part[(long long)b * 320];
    float* gr = a.grad + (long long)g * a.P;
    if (tap < 288) gr[OFF_W1C + tap] = s;
    else gr[OFF_B1C + tap - 288] = s;
  }
}

// optimizer update: vector grid-stride over (g, p/4) with a scalar
// tail. The SGD/Adam branch is hoisted OUT of the element loop and the
// state rows use alignment-4 float4s (rows are only dword-aligned at
// odd P), so the compiler emits merged dwordx4 traffic instead of the
// 28 scalar loads the per-element branch produced.
typedef float f4u __attribute__((ext_vector_type(4), aligned(4)));

extern "C" __global__ __launch_bounds__(WG)
void cnn_opt_step(CnnArgs a) {
  const float b1 = 0.9f, b2 = 0.999f, eps = 1e-8f;
  const long long pv = a.P / 4;          // float4 body; tail scalar
  const long long total = (long long)a.G * pv;
  for (long long qv = (long long)blockIdx.x * WG + threadIdx.x;
       qv < total; qv += (long long)gridDim.x * WG) {
    const int g = (int)(qv / pv);
    if (step_n(a, g) == 0) continue;
    const long long row = a.rows[g];
    const float lr_ = a.lr[row];
    const long long pp = (qv - (long long)g * pv) * 4;
    const long long q0 = (long long)g * a.P + pp;
    const bool tail = (qv - (long long)g * pv == pv - 1);
    if (a.opt == OPT_SGD) {
      f4u wv = *(const f4u*)(a.work + q0);
      const f4u gr = *(const f4u*)(a.grad + q0);
#pragma unroll
      for (int j = 0; j < 4; ++j) wv[j] -= lr_ * gr[j];
      *(f4u*)(a.work + q0) = wv;
      if (tail)
        for (long long q = q0 + 4; q < (long long)(g + 1) * a.P; ++q)
          a.work[q] -= lr_ * a.grad[q];
      continue;
    }
    const int tnew = a.t[row] + 1;       // tick kernel commits after
    const float bc1 = 1.f - powf(b1, (float)tnew);
    const float bc2 = 1.f - powf(b2, (float)tnew);
    const long long s0 = row * a.P + pp;
    f4u wv = *(const f4u*)(a.work + q0);
    const f4u gr0 = *(const f4u*)(a.grad + q0);
    f4u mo = *(const f4u*)(a.m + s0);
    f4u vo = *(const f4u*)(a.v + s0);
    f4u vm = *(const f4u*)(a.vmax + s0);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float gr = gr0[j] + a.wd * wv[j];
      const float mn = b1 * mo[j] + (1.f - b1) * gr;
      const float vn = b2 * vo[j] + (1.f - b2) * gr * gr;
      mo[j] = mn;
      vo[j] = vn;
      const float v2 = fmaxf(vm[j], vn);
      vm[j] = v2;
      wv[j] -= lr_ * (mn / bc1) / (sqrtf(v2 / bc2) + eps);
    }
    *(f4u*)(a.m + s0) = mo;
    *(f4u*)(a.v + s0) = vo;
    *(f4u*)(a.vmax + s0) = vm;
    *(f4u*)(a.work + q0) = wv;
    if (tail) {
      for (long long q = q0 + 4; q < (long long)(g + 1) * a.P; ++q) {
        const long long gp = row * a.P + (q - (long long)g * a.P);
        const float gr = a.grad[q] + a.wd * a.work[q];
        const float mn = b1 * a.m[gp] + (1.f - b1) * gr;
        const float vn = b2 * a.v[gp] + (1.f - b2) * gr * gr;
        a.m[gp] = mn;
        a.v[gp] = vn;
        const float v2 = fmaxf(a.vmax[gp], vn);
        a.vmax[gp] = v2;
        a.work[q] -= lr_ * (mn / bc1) / (sqrtf(v2 / bc2) + eps);
      }
    }
  }
}

extern "C" __global__ void cnn_opt_tick(CnnArgs a) {
  const int g = blockIdx.x * blockDim.x + threadIdx.x;
  if (g < a.G && a.opt == OPT_ADAM && step_n(a, g) > 0)
    a.t[a.rows[g]] += 1;
}

// ---------------------------------------------------------------------------
// evaluation: conv stage (per-sample block, x1 in LDS) + fc stage
// (per-window block, amortizes the fc1 weight stream over the window)
// ---------------------------------------------------------------------------

#define EV_ACC 0
#define EV_CONF 1
#define EV_DUMP 2

struct CnnEvalArgs {
  const float* __restrict__ params;     // [M, P]
  const int64_t* __restrict__ task_row; // [W]
  const int64_t* __restrict__ task_id;  // [W]
  const int64_t* __restrict__ off;      // [W]
  const int64_t* __restrict__ len;      // [W]
  const int64_t* __restrict__ slot;     // [W] prefix base into a2e
  const float* __restrict__ x;
  const int64_t* __restrict__ y;
  const float* __restrict__ x_mask;     // [W, 784] or [784] or null
  int xm_per_task;
  float* __restrict__ a2e;              // [slots, NF]
  float* __restrict__ z1e;              // [slots, NH] (fc1 output)
  float* __restrict__ z1pe;             // [slots, KS, NH] fc1 K-split partials
  // fc1 GEMM block metadata (slots grouped by model row, <=64 per block)
  const int64_t* __restrict__ blk_row;
  const int64_t* __restrict__ blk_s0;
  const int64_t* __restrict__ blk_len;
  // per-slot metadata for the head kernel
  const int64_t* __restrict__ srow;     // model row per slot
  const int64_t* __restrict__ stid;     // task id per slot
  const int64_t* __restrict__ sy;       // label per slot
  const int64_t* __restrict__ soff;     // arena sample index per slot
  const int64_t* __restrict__ swin;     // window index per slot (masks)
  float* __restrict__ x1e;              // [slots, X1N] channels-last
  float* __restrict__ z2e;              // [slots, Z2N] channels-last
  const float* __restrict__ wtf_e;      // [M, 9, 32, 64] reshaped conv2 W
  long long n_slots;
  // mode EV_ACC
  double* __restrict__ correct;         // [T]
  double* __restrict__ total;
  double* __restrict__ loss;
  double* __restrict__ mse;             // or null
  // mode EV_CONF: conf [T, O, O]; mode EV_DUMP: outp [slots, O]
  double* __restrict__ conf;
  float* __restrict__ outp;
  int O, P, mode;
};

// eval conv1: block per slot; the masked input stages in LDS once and
// each thread produces channels-last x1 elements (9 fused MACs each)
extern "C" __global__ __launch_bounds__(WG)
void cnn_eval_conv1(CnnEvalArgs a) {
  const long long slot = blockIdx.x;
  if (slot >= a.n_slots) return;
  const int tid = threadIdx.x;
  __shared__ __attribute__((aligned(16))) float xin[D_IN];
  __shared__ __attribute__((aligned(16))) float wc[288 + C1];
  const float* wp = a.params + a.srow[slot] * (long long)a.P;
  const float* xs = a.x + a.soff[slot] * D_IN;
  const float* xm = a.x_mask
      ? a.x_mask + (a.xm_per_task ? a.swin[slot] * D_IN : 0) : nullptr;
  for (int d = tid; d < D_IN; d += WG)
    xin[d] = xm ? xs[d] * xm[d] : xs[d];
  for (int i = tid; i < 288 + C1; i += WG) wc[i] = wp[OFF_W1C + i];
  __syncthreads();
  float* out = a.x1e + slot * X1N;
  for (int e = tid; e < X1N; e += WG) {
    const int pp = e / C1;
    const int c = e - pp * C1;
    const int oy = pp / S1, ox = pp - (pp / S1) * S1;
    float z = wc[288 + c];
#pragma unroll
    for (int ky = 0; ky < 3; ++ky)
#pragma unroll
      for (int kx = 0; kx < 3; ++kx)
        z = fmaf(xin[(oy + ky) * IN_W + ox + kx],
                 wc[c * 9 + ky * 3 + kx], z);
    out[e] = z;
  }
}

// eval conv2 as the region-staged MFMA GEMM (same structure as the
// train kernel, indexed by slot with per-slot model rows)
extern "C" __global__ __launch_bounds__(WG)
void cnn_eval_conv2_mfma(CnnEvalArgs a) {
  const int pt = blockIdx.x % 9;
  const long long slot = blockIdx.x / 9;
  if (slot >= a.n_slots) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  __shared__ __attribute__((aligned(16))) float sR[6 * S1][C1 + 1];
  const int p0 = pt * 64;
  const int r0 = p0 / S2;
  const float* x1 = a.x1e + slot * X1N + (long long)r0 * S1 * C1;
  {
    const int nrow = min(6 * S1, (S1 - r0) * S1);
    const int r8 = tid >> 5, kk = tid & 31;
    for (int rr = r8; rr < 6 * S1; rr += 8)
      sR[rr][kk] = (rr < nrow) ? x1[(long long)rr * C1 + kk] : 0.f;
  }
  __syncthreads();
  f32x4 acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};
  const int p = p0 + wv * 16 + li;
  const int arow = (p / S2 - r0) * S1 + (p - (p / S2) * S2);
  const long long row = a.srow[slot];
  const float* wt = a.wtf_e + row * (9 * 2048);
  for (int kyx = 0; kyx < 9; ++kyx) {
    const int ky = kyx / 3, kx = kyx - (kyx / 3) * 3;
    const int off = arow + ky * S1 + kx;
    const float* wk = wt + kyx * 2048;
#pragma unroll
    for (int ks = 0; ks < C1 / 4; ++ks) {
      const float av = sR[off][ks * 4 + lk];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
        acc[ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            av, wk[(ks * 4 + lk) * C2 + ct * 16 + li], acc[ct], 0, 0, 0);
    }
  }
  const float* bias = a.params + row * (long long)a.P + OFF_B2C;
  float* z2 = a.z2e + slot * Z2N;
#pragma unroll
  for (int ct = 0; ct < 4; ++ct) {
    const int co = ct * 16 + li;
    const float bb = bias[co];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int pp = p0 + wv * 16 + lk * 4 + r;
      z2[(long long)pp * C2 + co] = acc[ct][r] + bb;
    }
  }
}

// eval conv2, 128-pixel tiles (see cnn_conv2_fwd_mfma2)
extern "C" __global__ __launch_bounds__(WG)
void cnn_eval_conv2_mfma2(CnnEvalArgs a) {
  const int pt = blockIdx.x % 5;
  const long long slot = blockIdx.x / 5;
  if (slot >= a.n_slots) return;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15, lk = l >> 4;
  __shared__ __attribute__((aligned(16))) float sR[8 * S1][C1 + 1];
  const int p0 = pt * 128;
  const int r0 = p0 / S2;
  const float* x1 = a.x1e + slot * X1N + (long long)r0 * S1 * C1;
  {
    const int nrow = min(8 * S1, (S1 - r0) * S1);
    const int r8 = tid >> 5, kk = tid & 31;
    for (int rr = r8; rr < 8 * S1; rr += 8)
      sR[rr][kk] = (rr < nrow) ? x1[(long long)rr * C1 + kk] : 0.f;
  }
  __syncthreads();
  f32x4 acc[2][4];
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int t = 0; t < 4; ++t) acc[mf][t] = {0.f, 0.f, 0.f, 0.f};
  const int p_a = p0 + wv * 32 + li;
  const int p_b = p_a + 16;
  const int arow_a = (p_a / S2 - r0) * S1 + (p_a - (p_a / S2) * S2);
  const int arow_b = (p_b / S2 - r0) * S1 + (p_b - (p_b / S2) * S2);
  const long long row = a.srow[slot];
  const float* wt = a.wtf_e + row * (9 * 2048);
  for (int kyx = 0; kyx < 9; ++kyx) {
    const int ky = kyx / 3, kx = kyx - (kyx / 3) * 3;
    const int offa = arow_a + ky * S1 + kx;
    const int offb = arow_b + ky * S1 + kx;
    const float* wk = wt + kyx * 2048;
#pragma unroll
    for (int ks = 0; ks < C1 / 4; ++ks) {
      const float av0 = sR[offa][ks * 4 + lk];
      const float av1 = sR[offb][ks * 4 + lk];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        const float wv_ = wk[(ks * 4 + lk) * C2 + ct * 16 + li];
        acc[0][ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(av0, wv_,
                                                          acc[0][ct],
                                                          0, 0, 0);
        acc[1][ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(av1, wv_,
                                                          acc[1][ct],
                                                          0, 0, 0);
      }
    }
  }
  const float* bias = a.params + row * (long long)a.P + OFF_B2C;
  float* z2 = a.z2e + slot * Z2N;
#pragma unroll
  for (int ct = 0; ct < 4; ++ct) {
    const int co = ct * 16 + li;
    const float bb = bias[co];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int pp = p0 + wv * 32 + mf * 16 + lk * 4 + r;
        if (pp < Z2N / C2)
          z2[(long long)pp * C2 + co] = acc[mf][ct][r] + bb;
      }
  }
}

// eval maxpool, block per slot: channels-last z2 reads coalesced via
// LDS row staging, pooled a2e assembled in LDS and stored in one
// coalesced pass (torch-flatten layout; no dropout, no argmax at eval)
extern "C" __global__ __launch_bounds__(WG)
void cnn_eval_pool(CnnEvalArgs a) {
  const long long slot = blockIdx.x;
  if (slot >= a.n_slots) return;
  const int tid = threadIdx.x;
  __shared__ __attribute__((aligned(16))) float srow_[2][2 * S2 * C2];
  __shared__ __attribute__((aligned(16))) float sa2[NF];
  const float* z2 = a.z2e + slot * Z2N;
  // double-buffered row-pair staging (same pattern as cnn_pool_fwd):
  // the next pooled row's z2 loads issue before this row's max pass
  float rst[12];
#define PE_LOAD(py)                                                     \
  _Pragma("unroll") for (int jj = 0; jj < 12; ++jj)                      \
    rst[jj] = z2[(long long)(2 * (py)) * S2 * C2 + tid + jj * WG];
#define PE_WRITE(buf)                                                   \
  _Pragma("unroll") for (int jj = 0; jj < 12; ++jj)                      \
    srow_[buf][tid + jj * WG] = rst[jj];
  PE_LOAD(0);
  PE_WRITE(0);
  __syncthreads();
  int cur = 0;
  for (int py = 0; py < SP; ++py) {
    if (py + 1 < SP) { PE_LOAD(py + 1); }
    for (int q = tid; q < C2 * SP; q += WG) {
      const int c = q / SP;
      const int px = q - c * SP;
      float best = -1e30f;
#pragma unroll
      for (int dy = 0; dy < 2; ++dy)
#pragma unroll
        for (int dx = 0; dx < 2; ++dx)
          best = fmaxf(best, srow_[cur][(dy * S2 + 2 * px + dx) * C2 + c]);
      sa2[c * (SP * SP) + py * SP + px] = best;
    }
    if (py + 1 < SP) { PE_WRITE(cur ^ 1); }
    __syncthreads();
    cur ^= 1;
  }
#undef PE_LOAD
#undef PE_WRITE
  float* out = a.a2e + slot * (long long)NF;
  for (int e = tid; e < NF; e += WG) out[e] = sa2[e];
}

// fc1 eval as an MFMA tile GEMM with the SAME K-split as the training
// fc1 (the eval grid is otherwise tiny: ~4000 slots = 63 blocks for
// 256 CUs — per-wave PMC showed healthy 37% MFMA busy but only a
// quarter of the chip occupied): block (blk, ks) computes the partial
// z1 for up to 64 consecutive same-row slots x 128 h over K-range
// [ks*NF/KS, ...), partials land in z1pe and cnn_eval_fc1_act folds
// them with bias + relu. 4 waves x 16 rows x 128 cols, f32-input MFMA,
// BK=32 double-buffered register-staged LDS tiles with +1 padding.
#define EVAL_FC1_KS 8
extern "C" __global__ __launch_bounds__(WG)
void cnn_eval_fc1_mfma(CnnEvalArgs a) {
  const int ks = blockIdx.x % EVAL_FC1_KS;
  const int blk = blockIdx.x / EVAL_FC1_KS;
  const long long row = a.blk_row[blk];
  const long long s0 = a.blk_s0[blk];
  const int mlen = (int)a.blk_len[blk];
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int l = tid & 63;
  const int li = l & 15;
  const int lk = l >> 4;
  __shared__ __attribute__((aligned(16))) float sA[2][64][EVAL_BK + 1];
  __shared__ __attribute__((aligned(16))) float sB[2][EVAL_BK][NH + 1];
  f32x4 acc[8];
#pragma unroll
  for (int t = 0; t < 8; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};
  const float* wp = a.params + row * (long long)a.P + OFF_W1F;
  const int r8 = tid >> 5, kk = tid & 31;
  const int k_lo = ks * (NF / EVAL_FC1_KS);
  const int k_hi = (ks + 1) * (NF / EVAL_FC1_KS);
  // register staging: 8 A rows + 16 B rows per thread per tile
  float ra[8], rb[16];
#define EV_LOAD(k0)                                                     \
  _Pragma("unroll") for (int j = 0; j < 8; ++j) {                        \
    const int rr = r8 + j * 8;                                           \
    ra[j] = (rr < mlen) ? a.a2e[(s0 + rr) * (long long)NF + (k0) + kk]   \
                        : 0.f;                                           \
  }                                                                      \
  _Pragma("unroll") for (int j = 0; j < 16; ++j)                         \
    rb[j] = wp[(long long)(r8 + j * 8) * NF + (k0) + kk];
#define EV_WRITE(buf)                                                   \
  _Pragma("unroll") for (int j = 0; j < 8; ++j)                          \
    sA[buf][r8 + j * 8][kk] = ra[j];                                     \
  _Pragma("unroll") for (int j = 0; j < 16; ++j)                         \
    sB[buf][kk][r8 + j * 8] = rb[j];
  EV_LOAD(k_lo);
  EV_WRITE(0);
  __syncthreads();
  int cur = 0;
  for (int k0 = k_lo; k0 < k_hi; k0 += EVAL_BK) {
    if (k0 + EVAL_BK < k_hi) { EV_LOAD(k0 + EVAL_BK); }
#pragma unroll
    for (int kq = 0; kq < EVAL_BK / 4; ++kq) {
      const float av = sA[cur][wv * 16 + li][kq * 4 + lk];
#pragma unroll
      for (int ct = 0; ct < 8; ++ct)
        acc[ct] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            av, sB[cur][kq * 4 + lk][ct * 16 + li], acc[ct], 0, 0, 0);
    }
    if (k0 + EVAL_BK < k_hi) { EV_WRITE(cur ^ 1); }
    __syncthreads();
    cur ^= 1;
  }
#undef EV_LOAD
#undef EV_WRITE
  // epilogue: partials out (bias + relu fold in cnn_eval_fc1_act)
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mrow = wv * 16 + lk * 4 + r;
      if (mrow < mlen) {
        const int h = ct * 16 + li;
        a.z1pe[((s0 + mrow) * EVAL_FC1_KS + ks) * (long long)NH + h] =
            acc[ct][r];
      }
    }
  }
}

// fold the eval fc1 K-split partials + bias, apply relu -> z1e
extern "C" __global__ __launch_bounds__(WG)
void cnn_eval_fc1_act(CnnEvalArgs a) {
  const long long total = a.n_slots * NH;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const long long slot = q / NH;
    const int h = (int)(q - slot * NH);
    const float* wp = a.params + a.srow[slot] * (long long)a.P;
    float z = wp[OFF_B1F + h];
#pragma unroll
    for (int ks = 0; ks < EVAL_FC1_KS; ++ks)
      z += a.z1pe[(slot * EVAL_FC1_KS + ks) * (long long)NH + h];
    a.z1e[q] = z > 0.f ? z : 0.f;
  }
}

// head: per-slot fc2 + in-graph softmax + mode tail. ONE WAVE per slot
// (same shape as cnn_head_fwd): lane o owns output o, the z1 row loads
// once into two coalesced registers, softmax/argmax run as shuffle
// reductions, atomic tails issue from lane 0.
extern "C" __global__ __launch_bounds__(WG)
void cnn_eval_head(CnnEvalArgs a) {
  const int l = threadIdx.x & 63;
  for (long long slot = (long long)blockIdx.x * (WG / 64)
                        + (threadIdx.x >> 6);
       slot < a.n_slots; slot += (long long)gridDim.x * (WG / 64)) {
    const long long row = a.srow[slot];
    const float* wp = a.params + row * (long long)a.P;
    const float* z1 = a.z1e + slot * NH;
    const float r0 = z1[l], r1 = z1[64 + l];
    float zlane = 0.f;
    for (int o = 0; o < a.O; ++o) {
      const float* wo = wp + OFF_W2F + (long long)o * NH;
      const float v = wave_sum(fmaf(r0, wo[l], r1 * wo[64 + l]));
      if (l == o) zlane = v;
    }
    const bool live = l < a.O;
    float z = live ? zlane + wp[OFF_W2F + (long long)a.O * NH + l]
                   : -1e30f;
    const float zmax = wave_max(z);
    float e = live ? __expf(z - zmax) : 0.f;
    const float s = e / wave_sum(e);   // the model OUTPUT (in-graph softmax)
    // argmax, LOWEST index on ties (the sequential scan used strict >)
    float bv = live ? s : -1e30f;
    int bi = live ? l : 64;
#pragma unroll
    for (int m = 1; m < 64; m <<= 1) {
      const float ov = __shfl_xor(bv, m, 64);
      const int oi = __shfl_xor(bi, m, 64);
      if (ov > bv || (ov == bv && oi < bi)) { bv = ov; bi = oi; }
    }
    const int best = bi;
    const long long tsk = a.stid[slot];
    const int yi = (int)a.sy[slot];
    if (a.mode == EV_DUMP) {
      if (live) a.outp[slot * (long long)a.O + l] = s;
    } else if (a.mode == EV_CONF) {
      if (l == 0) atomicAdd(&a.conf[(tsk * a.O + yi) * a.O + best], 1.0);
    } else {
      // CE / mse on the softmax OUTPUT (double-softmax quirk)
      const float smax = wave_max(live ? s : -1e30f);
      const float ssum = wave_sum(live ? __expf(s - smax) : 0.f);
      const float lse = logf(ssum) + smax;
      const float syi = __shfl(s, yi, 64);
      if (l == 0) {
        atomicAdd(&a.correct[tsk], (double)((best == yi) ? 1.f : 0.f));
        atomicAdd(&a.total[tsk], 1.0);
        atomicAdd(&a.loss[tsk], (double)(lse - syi));
        if (a.mse) {
          const float pt = __expf(syi - lse);
          atomicAdd(&a.mse[tsk], (double)((1.f - pt) * (1.f - pt)));
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host side
// ---------------------------------------------------------------------------

// conv2 fwd/eval variant: 64-pixel tiles (1) vs 128-pixel shared-load
// tiles (2); FEDDRIFT_CONV2FWD overrides
static bool fwd_v2() {
  static const int v = [] {
    const char* e = getenv("FEDDRIFT_CONV2FWD");
    return e ? atoi(e) : 1;
  }();
  return v == 2;
}

static int grid_for(long long total) {
  long long b = (total + WG - 1) / WG;
  if (b > 16384) b = 16384;          // grid-stride beyond
  return (int)(b < 1 ? 1 : b);
}

void cnn_train_epoch_impl(
    torch::Tensor work, torch::Tensor grad, torch::Tensor rows,
    torch::Tensor x, torch::Tensor y,
    torch::Tensor step_off, torch::Tensor step_len, int64_t e,
    c10::optional<torch::Tensor> x_mask,
    torch::Tensor ws_x1, torch::Tensor ws_a2, torch::Tensor ws_pidx,
    torch::Tensor ws_z1, torch::Tensor ws_a1, torch::Tensor ws_dz2,
    torch::Tensor ws_dz1, torch::Tensor ws_da2, torch::Tensor ws_zz2,
    torch::Tensor ws_dx1, torch::Tensor ws_c1part, torch::Tensor ws_wtf,
    torch::Tensor ws_wtd, torch::Tensor ws_z1part, torch::Tensor ws_w2part,
    torch::Tensor ws_b2part,
    c10::optional<torch::Tensor> m, c10::optional<torch::Tensor> v,
    c10::optional<torch::Tensor> vmax, c10::optional<torch::Tensor> t,
    torch::Tensor lr, double wd, double p1, double p2,
    int64_t seed, int64_t g0, int64_t B, int64_t O, int64_t w2ms) {
  const int G = rows.size(0);
  if (G == 0) return;
  CnnArgs a;
  a.work = work.data_ptr<float>();
  a.grad = grad.data_ptr<float>();
  a.rows = rows.data_ptr<int64_t>();
  a.x = x.data_ptr<float>();
  a.y = y.data_ptr<int64_t>();
  a.step_off = step_off.data_ptr<int64_t>();
  a.step_len = step_len.data_ptr<int64_t>();
  a.x_mask = x_mask.has_value() ? x_mask->data_ptr<float>() : nullptr;
  a.x1 = ws_x1.data_ptr<float>();
  a.a2 = ws_a2.data_ptr<float>();
  a.pidx = ws_pidx.data_ptr<unsigned char>();
  a.z1 = ws_z1.data_ptr<float>();
  a.a1 = ws_a1.data_ptr<float>();
  a.dz2 = ws_dz2.data_ptr<float>();
  a.dz1 = ws_dz1.data_ptr<float>();
  a.da2 = ws_da2.data_ptr<float>();
  a.zz2 = ws_zz2.data_ptr<float>();
  a.dx1 = ws_dx1.data_ptr<float>();
  a.c1part = ws_c1part.data_ptr<float>();
  a.wtf = ws_wtf.data_ptr<float>();
  a.wtd = ws_wtd.data_ptr<float>();
  a.z1part = ws_z1part.data_ptr<float>();
  a.w2part = ws_w2part.data_ptr<float>();
  a.b2part = ws_b2part.data_ptr<float>();
  const bool adam = m.has_value();
  a.m = adam ? m->data_ptr<float>() : nullptr;
  a.v = adam ? v->data_ptr<float>() : nullptr;
  a.vmax = adam ? vmax->data_ptr<float>() : nullptr;
  a.t = adam ? t->data_ptr<int>() : nullptr;
  a.lr = lr.data_ptr<float>();
  a.wd = (float)wd;
  a.p1 = (float)p1;
  a.p2 = (float)p2;
  a.seed = (unsigned long long)seed;
  a.g0 = g0;
  a.G = G;
  a.B = (int)B;
  a.E = (int)step_off.size(1);
  a.e = (int)e;
  a.O = (int)O;
  a.P = (int)work.size(1);
  a.opt = adam ? OPT_ADAM : OPT_SGD;
  a.w2ms = (int)w2ms;

  auto s = c10::hip::getCurrentHIPStream();
  const long long GB = (long long)G * B;
#define L(kern, total) \
  hipLaunchKernelGGL(kern, dim3(grid_for(total)), dim3(WG), 0, s, a)
  const int mtiles = (a.B + 63) / 64;
  // forward
  L(cnn_w2_reshape, (long long)G * C2 * C1 * 9);
  hipLaunchKernelGGL(cnn_conv1_fwd, dim3((int)GB), dim3(WG), 0, s, a);
  if (fwd_v2())
    hipLaunchKernelGGL(cnn_conv2_fwd_mfma2, dim3((int)GB * 5), dim3(WG), 0,
                       s, a);
  else
    hipLaunchKernelGGL(cnn_conv2_fwd_mfma, dim3((int)GB * 9), dim3(WG), 0,
                       s, a);
  hipLaunchKernelGGL(cnn_pool_fwd, dim3((int)GB), dim3(WG), 0, s, a);
  hipLaunchKernelGGL(cnn_fc1_fwd_mfma, dim3(G * mtiles * FC1_KS), dim3(WG),
                     0, s, a);
  L(cnn_fc1_act, GB * NH);
  L(cnn_head_fwd, GB * 64);   // one wave per sample
  // backward
  L(cnn_fc2_wgrad, (long long)G * O * NH);
  L(cnn_fc2_dgrad, GB * NH);
  hipLaunchKernelGGL(cnn_fc1_wgrad, dim3(G * (NF / 128)), dim3(WG), 0, s,
                     a);
  L(cnn_fc1_bias_grad, (long long)G * NH);
  hipLaunchKernelGGL(cnn_fc1_dgrad, dim3(G * mtiles * (NF / 128)),
                     dim3(WG), 0, s, a);
  hipLaunchKernelGGL(cnn_pool_bwd, dim3((int)GB), dim3(WG), 0, s, a);
  hipLaunchKernelGGL(cnn_conv2_wgrad_mfma, dim3(G * a.w2ms), dim3(WG),
                     0, s, a);
  L(cnn_conv2_wgrad_reduce, (long long)G * (9 * 2048 + C2));
  // dgrad variant switch, A/B-measured at both ends of the fleet-size
  // axis (FEDDRIFT_DGRAD=1|2|3 forces one): v3 (128-pixel tiles, every
  // weight load shared by two M-fragments, 4 MFMA chains/wave) matches
  // v2 at small fleets (config-3 train 8.0 ms/round both) and beats
  // both the fwd-shaped v2 (5.97) and the all-LDS glds pipeline v1
  // (5.6-5.7) at 3400-client scale: 5.2 s/round -> default everywhere.
  static const int dgrad_env = [] {
    const char* e = getenv("FEDDRIFT_DGRAD");
    return e ? atoi(e) : 0;
  }();
  const int dgrad_v = dgrad_env ? dgrad_env : 3;
  if (dgrad_v == 1)
    hipLaunchKernelGGL(cnn_conv2_dgrad_mfma, dim3((int)GB), dim3(WG),
                       (9 * 2048 + 2 * DG_RC * 64) * sizeof(float), s, a);
  else if (dgrad_v == 3)
    hipLaunchKernelGGL(cnn_conv2_dgrad_mfma3, dim3((int)GB * 6), dim3(WG),
                       0, s, a);
  else
    hipLaunchKernelGGL(cnn_conv2_dgrad_mfma2, dim3((int)GB * 11), dim3(WG),
                       0, s, a);
  hipLaunchKernelGGL(cnn_conv1_wgrad_part, dim3(G * (int)B), dim3(WG), 0,
                     s, a);
  L(cnn_conv1_wgrad_reduce, (long long)G * 320);
  // optimizer
  L(cnn_opt_step, (long long)G * a.P);
  hipLaunchKernelGGL(cnn_opt_tick, dim3((G + WG - 1) / WG), dim3(WG), 0, s,
                     a);
#undef L
  TORCH_CHECK(hipGetLastError() == hipSuccess, "cnn_train_epoch launch");
}

torch::Tensor cnn_eval(
    torch::Tensor params, torch::Tensor task_row, torch::Tensor task_id,
    torch::Tensor off, torch::Tensor len, torch::Tensor slot,
    torch::Tensor x, torch::Tensor y, torch::Tensor a2e, torch::Tensor z1e,
    torch::Tensor z1pe,
    torch::Tensor x1e, torch::Tensor z2e, torch::Tensor wtf_e,
    torch::Tensor blk_row, torch::Tensor blk_s0, torch::Tensor blk_len,
    torch::Tensor srow, torch::Tensor stid, torch::Tensor sy,
    torch::Tensor soff, torch::Tensor swin,
    c10::optional<torch::Tensor> x_mask, int64_t n_tasks, int64_t O,
    int64_t mode, bool want_mse, int64_t max_len, int64_t n_slots,
    c10::optional<torch::Tensor> outp) {
  const int W = task_row.size(0);
  auto optd = torch::TensorOptions().dtype(torch::kFloat64)
                  .device(params.device());
  torch::Tensor out;
  if (mode == EV_ACC)
    out = torch::zeros({want_mse ? 4 : 3, n_tasks}, optd);
  else if (mode == EV_CONF)
    out = torch::zeros({n_tasks, O, O}, optd);
  else
    out = torch::zeros({0}, optd);
  if (W == 0 || n_slots == 0) return out;

  CnnEvalArgs a;
  a.params = params.data_ptr<float>();
  a.task_row = task_row.data_ptr<int64_t>();
  a.task_id = task_id.data_ptr<int64_t>();
  a.off = off.data_ptr<int64_t>();
  a.len = len.data_ptr<int64_t>();
  a.slot = slot.data_ptr<int64_t>();
  a.x = x.data_ptr<float>();
  a.y = y.data_ptr<int64_t>();
  a.x_mask = x_mask.has_value() ? x_mask->data_ptr<float>() : nullptr;
  a.xm_per_task = x_mask.has_value() && x_mask->dim() == 2 ? 1 : 0;
  a.a2e = a2e.data_ptr<float>();
  a.z1e = z1e.data_ptr<float>();
  a.z1pe = z1pe.data_ptr<float>();
  a.blk_row = blk_row.data_ptr<int64_t>();
  a.blk_s0 = blk_s0.data_ptr<int64_t>();
  a.blk_len = blk_len.data_ptr<int64_t>();
  a.srow = srow.data_ptr<int64_t>();
  a.stid = stid.data_ptr<int64_t>();
  a.sy = sy.data_ptr<int64_t>();
  a.soff = soff.data_ptr<int64_t>();
  a.swin = swin.data_ptr<int64_t>();
  a.x1e = x1e.data_ptr<float>();
  a.z2e = z2e.data_ptr<float>();
  a.wtf_e = wtf_e.data_ptr<float>();
  a.n_slots = (long long)n_slots;
  double* base = (mode == EV_ACC) ? out.data_ptr<double>() : nullptr;
  a.correct = base;
  a.total = base ? base + n_tasks : nullptr;
  a.loss = base ? base + 2 * n_tasks : nullptr;
  a.mse = (base && want_mse) ? base + 3 * n_tasks : nullptr;
  a.conf = (mode == EV_CONF) ? out.data_ptr<double>() : nullptr;
  a.outp = outp.has_value() ? outp->data_ptr<float>() : nullptr;
  a.O = (int)O;
  a.P = (int)params.size(1);
  a.mode = (int)mode;

  auto s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(cnn_eval_conv1, dim3((int)n_slots), dim3(WG), 0, s,
                     a);
  if (fwd_v2())
    hipLaunchKernelGGL(cnn_eval_conv2_mfma2, dim3((int)(n_slots * 5)),
                       dim3(WG), 0, s, a);
  else
    hipLaunchKernelGGL(cnn_eval_conv2_mfma, dim3((int)(n_slots * 9)),
                       dim3(WG), 0, s, a);
  hipLaunchKernelGGL(cnn_eval_pool, dim3((int)n_slots), dim3(WG), 0, s, a);
  hipLaunchKernelGGL(cnn_eval_fc1_mfma,
                     dim3((int)blk_row.size(0) * EVAL_FC1_KS),
                     dim3(WG), 0, s, a);
  hipLaunchKernelGGL(cnn_eval_fc1_act, dim3(grid_for(n_slots * NH)),
                     dim3(WG), 0, s, a);
  hipLaunchKernelGGL(cnn_eval_head, dim3(grid_for(n_slots * 64)), dim3(WG), 0,
                     s, a);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "cnn_eval launch");
  (void)max_len;
  return out;
}

void register_cnn(pybind11::module_& mod) {
  mod.def("cnn_train_epoch", &cnn_train_epoch_impl,
          "one fused CNN training epoch over all (client, model) pairs");
  mod.def("cnn_eval", &cnn_eval, "batched CNN eval sweep (acc/conf/dump)");
}
