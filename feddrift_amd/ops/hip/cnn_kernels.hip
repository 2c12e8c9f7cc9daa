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
  float* __restrict__ dz2c;          // [G, B, Z2N]
  float* __restrict__ dx1;           // [G, B, X1N]
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
  int G, B, E, e, O, P, opt;
};

#define OPT_SGD 0
#define OPT_ADAM 1

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

// conv1: one thread per (g, b, c1, y, x) output element, grid-stride
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv1_fwd(CnnArgs a) {
  const long long total = (long long)a.G * a.B * X1N;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)a.B * X1N));
    const long long r = q - (long long)g * a.B * X1N;
    const int b = (int)(r / X1N);
    if (b >= step_n(a, g)) continue;
    const int e = (int)(r - (long long)b * X1N);
    const int c = e / (S1 * S1);
    const int p = e - c * S1 * S1;
    const int oy = p / S1, ox = p - (p / S1) * S1;
    const float* w = a.work + (long long)g * a.P;
    const float* xs = a.x + (step_o(a, g) + b) * D_IN;
    const float* xm = a.x_mask ? a.x_mask + (long long)g * D_IN : nullptr;
    float z = w[OFF_B1C + c];
#pragma unroll
    for (int ky = 0; ky < 3; ++ky)
#pragma unroll
      for (int kx = 0; kx < 3; ++kx) {
        const int xi = (oy + ky) * IN_W + ox + kx;
        float xv = xs[xi];
        if (xm) xv *= xm[xi];
        z = fmaf(xv, w[OFF_W1C + c * 9 + ky * 3 + kx], z);
      }
    a.x1[q] = z;
  }
}

// conv2 + bias + maxpool + dropout1: thread per (g, b, c2, py, px)
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_pool_fwd(CnnArgs a) {
  const long long total = (long long)a.G * a.B * NF;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)a.B * NF));
    const long long r = q - (long long)g * a.B * NF;
    const int b = (int)(r / NF);
    if (b >= step_n(a, g)) continue;
    const int e = (int)(r - (long long)b * NF);
    const int c = e / (SP * SP);
    const int p = e - c * SP * SP;
    const int py = p / SP, px = p - (p / SP) * SP;
    const float* w = a.work + (long long)g * a.P;
    const float* x1 = a.x1 + ((long long)g * a.B + b) * X1N;
    float best = -1e30f;
    int arg = 0;
#pragma unroll
    for (int dy = 0; dy < 2; ++dy)
#pragma unroll
      for (int dx = 0; dx < 2; ++dx) {
        const int oy = 2 * py + dy, ox = 2 * px + dx;
        float z = w[OFF_B2C + c];
        for (int ci = 0; ci < C1; ++ci) {
          const float* xc = x1 + ci * S1 * S1 + oy * S1 + ox;
          const float* wc = w + OFF_W2C + (c * C1 + ci) * 9;
#pragma unroll
          for (int ky = 0; ky < 3; ++ky)
#pragma unroll
            for (int kx = 0; kx < 3; ++kx)
              z = fmaf(xc[ky * S1 + kx], wc[ky * 3 + kx], z);
        }
        if (z > best) { best = z; arg = dy * 2 + dx; }
      }
    a.pidx[q] = (unsigned char)arg;
    a.a2[q] = best * drop_scale(a, 0, g, b, e);
  }
}

// fc1 + relu + dropout2: thread per (g, b, h)
extern "C" __global__ __launch_bounds__(WG)
void cnn_fc1_fwd(CnnArgs a) {
  const long long total = (long long)a.G * a.B * NH;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)a.B * NH));
    const long long r = q - (long long)g * a.B * NH;
    const int b = (int)(r / NH);
    if (b >= step_n(a, g)) continue;
    const int h = (int)(r - (long long)b * NH);
    const float* w = a.work + (long long)g * a.P + OFF_W1F +
                     (long long)h * NF;
    const float* xin = a.a2 + ((long long)g * a.B + b) * NF;
    float z = a.work[(long long)g * a.P + OFF_B1F + h];
    for (int j = 0; j < NF; ++j) z = fmaf(xin[j], w[j], z);
    a.z1[q] = z;
    const float rl = z > 0.f ? z : 0.f;
    a.a1[q] = rl * drop_scale(a, 1, g, b, h);
  }
}

// fc2 + softmax (in-graph model output s) + CE-on-s gradient -> dz2
// dL/ds = (softmax(s) - onehot(y)) / n; dL/dz2 = s*(dL/ds - sum(dL/ds*s))
extern "C" __global__ __launch_bounds__(WG)
void cnn_head_fwd(CnnArgs a) {
  const long long total = (long long)a.G * a.B;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / a.B);
    const int b = (int)(q - (long long)g * a.B);
    const int n = step_n(a, g);
    if (b >= n) continue;
    const float inv_n = 1.f / (float)n;
    const float* w = a.work + (long long)g * a.P;
    const float* a1 = a.a1 + q * NH;
    const int yi = (int)a.y[step_o(a, g) + b];
    float z2[64], s[64];
    float zmax = -1e30f;
    for (int o = 0; o < a.O; ++o) {
      float z = w[OFF_W2F + (long long)a.O * NH + o];  // bias after W2f
      const float* wo = w + OFF_W2F + (long long)o * NH;
      for (int h = 0; h < NH; ++h) z = fmaf(a1[h], wo[h], z);
      z2[o] = z;
      zmax = fmaxf(zmax, z);
    }
    float zsum = 0.f;
    for (int o = 0; o < a.O; ++o) { s[o] = __expf(z2[o] - zmax); zsum += s[o]; }
    for (int o = 0; o < a.O; ++o) s[o] /= zsum;
    // CE(log_softmax(s), y): q2 = softmax(s)
    float smax = -1e30f;
    for (int o = 0; o < a.O; ++o) smax = fmaxf(smax, s[o]);
    float ssum = 0.f;
    for (int o = 0; o < a.O; ++o) ssum += __expf(s[o] - smax);
    float dot = 0.f;
    float ds[64];
    for (int o = 0; o < a.O; ++o) {
      float q2 = __expf(s[o] - smax) / ssum;
      if (o == yi) q2 -= 1.f;
      ds[o] = q2 * inv_n;
      dot += ds[o] * s[o];
    }
    for (int o = 0; o < a.O; ++o)
      a.dz2[q * a.O + o] = s[o] * (ds[o] - dot);
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

// fc1 wgrad: thread per (g, h, j), exclusive writes; bias at j == 0
extern "C" __global__ __launch_bounds__(WG)
void cnn_fc1_wgrad(CnnArgs a) {
  const long long total = (long long)a.G * NH * NF;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)NH * NF));
    const long long r = q - (long long)g * NH * NF;
    const int h = (int)(r / NF);
    const int j = (int)(r - (long long)h * NF);
    const int n = step_n(a, g);
    if (n == 0) continue;
    float acc = 0.f, accb = 0.f;
    for (int b = 0; b < n; ++b) {
      const float d = a.dz1[((long long)g * a.B + b) * NH + h];
      acc = fmaf(d, a.a2[((long long)g * a.B + b) * NF + j], acc);
      if (j == 0) accb += d;
    }
    float* gr = a.grad + (long long)g * a.P;
    gr[OFF_W1F + (long long)h * NF + j] = acc;
    if (j == 0) gr[OFF_B1F + h] = accb;
  }
}

// fc1 dgrad: da2[b, j] = sum_h dz1[b, h] * W1f[h, j]; thread per (g, b, j)
extern "C" __global__ __launch_bounds__(WG)
void cnn_fc1_dgrad(CnnArgs a) {
  const long long total = (long long)a.G * a.B * NF;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)a.B * NF));
    const long long r = q - (long long)g * a.B * NF;
    const int b = (int)(r / NF);
    if (b >= step_n(a, g)) continue;
    const int j = (int)(r - (long long)b * NF);
    const float* w = a.work + (long long)g * a.P + OFF_W1F;
    const float* d1 = a.dz1 + ((long long)g * a.B + b) * NH;
    float acc = 0.f;
    for (int h = 0; h < NH; ++h)
      acc = fmaf(d1[h], w[(long long)h * NF + j], acc);
    a.da2[q] = acc;
  }
}

// pool backward: route da2 (through dropout1) to the argmax position;
// the other 3 positions of the 2x2 cell are written zero (exclusive owner)
extern "C" __global__ __launch_bounds__(WG)
void cnn_pool_bwd(CnnArgs a) {
  const long long total = (long long)a.G * a.B * NF;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)a.B * NF));
    const long long r = q - (long long)g * a.B * NF;
    const int b = (int)(r / NF);
    if (b >= step_n(a, g)) continue;
    const int e = (int)(r - (long long)b * NF);
    const int c = e / (SP * SP);
    const int p = e - c * SP * SP;
    const int py = p / SP, px = p - (p / SP) * SP;
    const float dv = a.da2[q] * drop_scale(a, 0, g, b, e);
    const int arg = a.pidx[q];
    float* dzc = a.dz2c + ((long long)g * a.B + b) * Z2N + c * S2 * S2;
#pragma unroll
    for (int dy = 0; dy < 2; ++dy)
#pragma unroll
      for (int dx = 0; dx < 2; ++dx)
        dzc[(2 * py + dy) * S2 + 2 * px + dx] =
            (dy * 2 + dx == arg) ? dv : 0.f;
  }
}

// conv2 wgrad: block per (g, c2); threads own the 288 (ci, ky, kx) taps
// and loop (b, pixels); bias via cooperative reduce
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_wgrad(CnnArgs a) {
  const int g = blockIdx.x / C2;
  const int c = blockIdx.x - g * C2;
  const int n = step_n(a, g);
  if (n == 0) return;
  const int tid = threadIdx.x;
  __shared__ float sdz[S2 * S2];
  __shared__ float sred[4];
  float acc[2] = {0.f, 0.f};  // up to 2 taps per thread (288 <= 2*256)
  float bacc = 0.f;
  for (int b = 0; b < n; ++b) {
    const float* dzc =
        a.dz2c + ((long long)g * a.B + b) * Z2N + c * S2 * S2;
    for (int p = tid; p < S2 * S2; p += WG) sdz[p] = dzc[p];
    __syncthreads();
    const float* x1 = a.x1 + ((long long)g * a.B + b) * X1N;
    for (int t = 0; t < 2; ++t) {
      const int tap = tid + t * WG;
      if (tap < 288) {
        const int ci = tap / 9;
        const int k = tap - ci * 9;
        const int ky = k / 3, kx = k - (k / 3) * 3;
        const float* xc = x1 + ci * S1 * S1 + ky * S1 + kx;
        float s = 0.f;
        for (int oy = 0; oy < S2; ++oy) {
          const float* xr = xc + oy * S1;
          const float* dr = sdz + oy * S2;
          for (int ox = 0; ox < S2; ++ox) s = fmaf(xr[ox], dr[ox], s);
        }
        acc[t] += s;
      }
    }
    // bias: cooperative sum of sdz
    float bs = 0.f;
    for (int p = tid; p < S2 * S2; p += WG) bs += sdz[p];
    bacc += bs;
    __syncthreads();
  }
  float* gr = a.grad + (long long)g * a.P;
  for (int t = 0; t < 2; ++t) {
    const int tap = tid + t * WG;
    if (tap < 288) gr[OFF_W2C + c * 288 + tap] = acc[t];
  }
  // block-reduce bacc
  for (int off = 32; off > 0; off >>= 1) bacc += __shfl_down(bacc, off, 64);
  const int lane = tid & 63, wave = tid >> 6;
  if (lane == 0) sred[wave] = bacc;
  __syncthreads();
  if (tid == 0)
    gr[OFF_B2C + c] = sred[0] + sred[1] + sred[2] + sred[3];
}

// conv2 dgrad: dx1[b, ci, y, x] = sum over valid (c2, ky, kx); thread per
// (g, b, ci, y, x)
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv2_dgrad(CnnArgs a) {
  const long long total = (long long)a.G * a.B * X1N;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / ((long long)a.B * X1N));
    const long long r = q - (long long)g * a.B * X1N;
    const int b = (int)(r / X1N);
    if (b >= step_n(a, g)) continue;
    const int e = (int)(r - (long long)b * X1N);
    const int ci = e / (S1 * S1);
    const int p = e - ci * S1 * S1;
    const int y = p / S1, x = p - (p / S1) * S1;
    const float* w = a.work + (long long)g * a.P + OFF_W2C;
    const float* dzc = a.dz2c + ((long long)g * a.B + b) * Z2N;
    float acc = 0.f;
#pragma unroll
    for (int ky = 0; ky < 3; ++ky) {
      const int oy = y - ky;
      if (oy < 0 || oy >= S2) continue;
#pragma unroll
      for (int kx = 0; kx < 3; ++kx) {
        const int ox = x - kx;
        if (ox < 0 || ox >= S2) continue;
        for (int co = 0; co < C2; ++co)
          acc = fmaf(dzc[co * S2 * S2 + oy * S2 + ox],
                     w[(co * C1 + ci) * 9 + ky * 3 + kx], acc);
      }
    }
    a.dx1[q] = acc;
  }
}

// conv1 wgrad: block per g; threads own the 288 + 32 grad entries
extern "C" __global__ __launch_bounds__(WG)
void cnn_conv1_wgrad(CnnArgs a) {
  const int g = blockIdx.x;
  const int n = step_n(a, g);
  if (n == 0) return;
  const int tid = threadIdx.x;
  float wacc[2] = {0.f, 0.f};
  float bacc = 0.f;
  for (int b = 0; b < n; ++b) {
    const float* xs = a.x + (step_o(a, g) + b) * D_IN;
    const float* xm = a.x_mask ? a.x_mask + (long long)g * D_IN : nullptr;
    const float* dx = a.dx1 + ((long long)g * a.B + b) * X1N;
    for (int t = 0; t < 2; ++t) {
      const int tap = tid + t * WG;
      if (tap < 288) {
        const int c = tap / 9;
        const int k = tap - c * 9;
        const int ky = k / 3, kx = k - (k / 3) * 3;
        const float* dc = dx + c * S1 * S1;
        float s = 0.f;
        for (int y = 0; y < S1; ++y)
          for (int x = 0; x < S1; ++x) {
            const int xi = (y + ky) * IN_W + x + kx;
            float xv = xs[xi];
            if (xm) xv *= xm[xi];
            s = fmaf(xv, dc[y * S1 + x], s);
          }
        wacc[t] += s;
      } else if (tap < 288 + C1) {
        const int c = tap - 288;
        const float* dc = dx + c * S1 * S1;
        float s = 0.f;
        for (int p = 0; p < S1 * S1; ++p) s += dc[p];
        bacc += s;
      }
    }
  }
  float* gr = a.grad + (long long)g * a.P;
  for (int t = 0; t < 2; ++t) {
    const int tap = tid + t * WG;
    if (tap < 288) gr[OFF_W1C + tap] = wacc[t];
    else if (tap < 288 + C1) gr[OFF_B1C + tap - 288] = bacc;
  }
}

// optimizer update: thread per (g, p) grid-stride; pairs with n == 0 skip
extern "C" __global__ __launch_bounds__(WG)
void cnn_opt_step(CnnArgs a) {
  const long long total = (long long)a.G * a.P;
  const float b1 = 0.9f, b2 = 0.999f, eps = 1e-8f;
  for (long long q = (long long)blockIdx.x * WG + threadIdx.x; q < total;
       q += (long long)gridDim.x * WG) {
    const int g = (int)(q / a.P);
    if (step_n(a, g) == 0) continue;
    const long long p = q - (long long)g * a.P;
    const long long row = a.rows[g];
    const float lr_ = a.lr[row];
    float wv = a.work[q];
    const float gr0 = a.grad[q];
    if (a.opt == OPT_SGD) {
      a.work[q] = wv - lr_ * gr0;
      continue;
    }
    const int tnew = a.t[row] + 1;  // tick kernel commits after
    // precise powf: __powf's fast-math error lands ~1e-5 off the torch
    // reference through the bias corrections (measured on the box)
    const float bc1 = 1.f - powf(b1, (float)tnew);
    const float bc2 = 1.f - powf(b2, (float)tnew);
    const long long gp = row * a.P + p;
    const float gr = gr0 + a.wd * wv;
    const float mn = b1 * a.m[gp] + (1.f - b1) * gr;
    const float vn = b2 * a.v[gp] + (1.f - b2) * gr * gr;
    a.m[gp] = mn;
    a.v[gp] = vn;
    const float vm = fmaxf(a.vmax[gp], vn);
    a.vmax[gp] = vm;
    a.work[q] = wv - lr_ * (mn / bc1) / (sqrtf(vm / bc2) + eps);
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

// per (window, sample) block: conv1 (LDS x1) + conv2 + pool -> a2e
extern "C" __global__ __launch_bounds__(WG)
void cnn_eval_conv(CnnEvalArgs a) {
  const int w = blockIdx.x;
  const int i = blockIdx.y;
  if (i >= (int)a.len[w]) return;
  const int tid = threadIdx.x;
  __shared__ __attribute__((aligned(16))) float xin[D_IN];
  extern __shared__ __attribute__((aligned(16))) float x1[];  // [X1N]
  const float* wp = a.params + a.task_row[w] * (long long)a.P;
  const float* xs = a.x + (a.off[w] + i) * D_IN;
  const float* xm = a.x_mask
      ? a.x_mask + (a.xm_per_task ? (long long)w * D_IN : 0) : nullptr;
  for (int d = tid; d < D_IN; d += WG)
    xin[d] = xm ? xs[d] * xm[d] : xs[d];
  __syncthreads();
  for (int e = tid; e < X1N; e += WG) {
    const int c = e / (S1 * S1);
    const int p = e - c * S1 * S1;
    const int oy = p / S1, ox = p - (p / S1) * S1;
    float z = wp[OFF_B1C + c];
#pragma unroll
    for (int ky = 0; ky < 3; ++ky)
#pragma unroll
      for (int kx = 0; kx < 3; ++kx)
        z = fmaf(xin[(oy + ky) * IN_W + ox + kx],
                 wp[OFF_W1C + c * 9 + ky * 3 + kx], z);
    x1[e] = z;
  }
  __syncthreads();
  float* out = a.a2e + (a.slot[w] + i) * (long long)NF;
  for (int e = tid; e < NF; e += WG) {
    const int c = e / (SP * SP);
    const int p = e - c * SP * SP;
    const int py = p / SP, px = p - (p / SP) * SP;
    float best = -1e30f;
#pragma unroll
    for (int dy = 0; dy < 2; ++dy)
#pragma unroll
      for (int dx = 0; dx < 2; ++dx) {
        const int oy = 2 * py + dy, ox = 2 * px + dx;
        float z = wp[OFF_B2C + c];
        for (int ci = 0; ci < C1; ++ci) {
          const float* xc = x1 + ci * S1 * S1 + oy * S1 + ox;
          const float* wc = wp + OFF_W2C + (c * C1 + ci) * 9;
#pragma unroll
          for (int ky = 0; ky < 3; ++ky)
#pragma unroll
            for (int kx = 0; kx < 3; ++kx)
              z = fmaf(xc[ky * S1 + kx], wc[ky * 3 + kx], z);
        }
        best = fmaxf(best, z);
      }
    out[e] = best;
  }
}

// per-window block (128 threads): fc1 + relu + fc2 + softmax + tail
extern "C" __global__ __launch_bounds__(NH)
void cnn_eval_fc(CnnEvalArgs a) {
  const int w = blockIdx.x;
  const int n = (int)a.len[w];
  if (n == 0) return;
  const int tid = threadIdx.x;
  const float* wp = a.params + a.task_row[w] * (long long)a.P;
  const long long tsk = a.task_id[w];
  __shared__ __attribute__((aligned(16))) float sa2[NF];
  __shared__ float sa1[NH];
  __shared__ float sz2[64];
  __shared__ float sred[2];
  float c_acc = 0.f, l_acc = 0.f, e_acc = 0.f;
  for (int i = 0; i < n; ++i) {
    const float* src = a.a2e + (a.slot[w] + i) * (long long)NF;
    for (int j = tid; j < NF; j += NH) sa2[j] = src[j];
    __syncthreads();
    // fc1: thread h computes z1[h]
    {
      const float* wr = wp + OFF_W1F + (long long)tid * NF;
      float z = wp[OFF_B1F + tid];
      for (int j = 0; j < NF; ++j) z = fmaf(sa2[j], wr[j], z);
      sa1[tid] = z > 0.f ? z : 0.f;
    }
    __syncthreads();
    // fc2: thread o < O
    if (tid < a.O) {
      const float* wo = wp + OFF_W2F + (long long)tid * NH;
      float z = wp[OFF_W2F + (long long)a.O * NH + tid];
      for (int h = 0; h < NH; ++h) z = fmaf(sa1[h], wo[h], z);
      sz2[tid] = z;
    }
    __syncthreads();
    if (tid == 0) {
      const int yi = (int)a.y[a.off[w] + i];
      float zmax = -1e30f;
      for (int o = 0; o < a.O; ++o) zmax = fmaxf(zmax, sz2[o]);
      float zsum = 0.f;
      float s[64];
      for (int o = 0; o < a.O; ++o) {
        s[o] = __expf(sz2[o] - zmax);
        zsum += s[o];
      }
      int best = 0;
      float bv = -1e30f;
      for (int o = 0; o < a.O; ++o) {
        s[o] /= zsum;  // the model OUTPUT (in-graph softmax)
        if (s[o] > bv) { bv = s[o]; best = o; }
      }
      if (a.mode == EV_DUMP) {
        for (int o = 0; o < a.O; ++o)
          a.outp[(a.slot[w] + i) * (long long)a.O + o] = s[o];
      } else if (a.mode == EV_CONF) {
        atomicAdd(&a.conf[(tsk * a.O + yi) * a.O + best], 1.0);
      } else {
        // CE / mse on the softmax OUTPUT (double-softmax quirk)
        float smax = -1e30f;
        for (int o = 0; o < a.O; ++o) smax = fmaxf(smax, s[o]);
        float ssum = 0.f;
        for (int o = 0; o < a.O; ++o) ssum += __expf(s[o] - smax);
        const float lse = logf(ssum) + smax;
        c_acc += (best == yi) ? 1.f : 0.f;
        l_acc += lse - s[yi];
        if (a.mse) {
          const float pt = __expf(s[yi] - lse);
          e_acc += (1.f - pt) * (1.f - pt);
        }
      }
    }
    __syncthreads();
  }
  if (a.mode == EV_ACC && tid == 0) {
    atomicAdd(&a.correct[tsk], (double)c_acc);
    atomicAdd(&a.total[tsk], (double)n);
    atomicAdd(&a.loss[tsk], (double)l_acc);
    if (a.mse) atomicAdd(&a.mse[tsk], (double)e_acc);
  }
  (void)sred;
}

// ---------------------------------------------------------------------------
// host side
// ---------------------------------------------------------------------------

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
    torch::Tensor ws_dz1, torch::Tensor ws_da2, torch::Tensor ws_dz2c,
    torch::Tensor ws_dx1,
    c10::optional<torch::Tensor> m, c10::optional<torch::Tensor> v,
    c10::optional<torch::Tensor> vmax, c10::optional<torch::Tensor> t,
    torch::Tensor lr, double wd, double p1, double p2,
    int64_t seed, int64_t g0, int64_t B, int64_t O) {
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
  a.dz2c = ws_dz2c.data_ptr<float>();
  a.dx1 = ws_dx1.data_ptr<float>();
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

  auto s = c10::hip::getCurrentHIPStream();
  const long long GB = (long long)G * B;
#define L(kern, total) \
  hipLaunchKernelGGL(kern, dim3(grid_for(total)), dim3(WG), 0, s, a)
  L(cnn_conv1_fwd, GB * X1N);
  L(cnn_conv2_pool_fwd, GB * NF);
  L(cnn_fc1_fwd, GB * NH);
  L(cnn_head_fwd, GB);
  L(cnn_fc2_wgrad, (long long)G * O * NH);
  L(cnn_fc2_dgrad, GB * NH);
  L(cnn_fc1_wgrad, (long long)G * NH * NF);
  L(cnn_fc1_dgrad, GB * NF);
  L(cnn_pool_bwd, GB * NF);
  hipLaunchKernelGGL(cnn_conv2_wgrad, dim3(G * C2), dim3(WG), 0, s, a);
  L(cnn_conv2_dgrad, GB * X1N);
  hipLaunchKernelGGL(cnn_conv1_wgrad, dim3(G), dim3(WG), 0, s, a);
  L(cnn_opt_step, (long long)G * a.P);
  hipLaunchKernelGGL(cnn_opt_tick, dim3((G + WG - 1) / WG), dim3(WG), 0, s,
                     a);
#undef L
  TORCH_CHECK(hipGetLastError() == hipSuccess, "cnn_train_epoch launch");
}

torch::Tensor cnn_eval(
    torch::Tensor params, torch::Tensor task_row, torch::Tensor task_id,
    torch::Tensor off, torch::Tensor len, torch::Tensor slot,
    torch::Tensor x, torch::Tensor y, torch::Tensor a2e,
    c10::optional<torch::Tensor> x_mask, int64_t n_tasks, int64_t O,
    int64_t mode, bool want_mse, int64_t max_len,
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
  if (W == 0) return out;

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
  hipLaunchKernelGGL(cnn_eval_conv, dim3(W, (int)max_len), dim3(WG),
                     X1N * sizeof(float), s, a);
  hipLaunchKernelGGL(cnn_eval_fc, dim3(W), dim3(NH), 0, s, a);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "cnn_eval launch");
  return out;
}

void register_cnn(pybind11::module_& mod) {
  mod.def("cnn_train_epoch", &cnn_train_epoch_impl,
          "one fused CNN training epoch over all (client, model) pairs");
  mod.def("cnn_eval", &cnn_eval, "batched CNN eval sweep (acc/conf/dump)");
}
