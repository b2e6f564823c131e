// CDNA4 (gfx950) kernels for the federated-learning hot ops.
//
// Design: every model's exchanged parameter set lives in ONE contiguous fp32
// flat buffer (fl4health_amd.parameter_exchange.FlatParameters). All per-round
// server math (weighted aggregation epilogues, server optimizers) and all
// per-step client math (proximal SGD, SCAFFOLD-corrected SGD, clipping, noise)
// are single fused kernels over that buffer: one launch, one pass over HBM,
// instead of the reference's per-layer NumPy loops
// (reference: fl4health/strategies/aggregate_utils.py:8-55,
//  fl4health/clients/scaffold_client.py:175-197,
//  fl4health/losses/weight_drift_loss.py:5-64,
//  fl4health/strategies/noisy_aggregate.py:7-122,
//  fl4health/strategies/flash.py:125-170 — all Python/NumPy there).
//
// These ops are HBM-bandwidth-bound elementwise/reduction work (no GEMM shape),
// so the CDNA4 mapping is: 256-thread blocks (4 wave64), float4 vectorized
// grid-stride loops sized to cover all 256 CUs across the 8 XCDs many times
// over, and deterministic two-stage reductions (fixed partial count) so
// aggregation results are bitwise reproducible run-to-run.
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <math.h>
#include <algorithm>
#include "philox.h"

#define BLOCK 256
// >=8x256 workgroups fills all 8 XCDs; grid-stride covers any n.
static inline int grid_1d(int64_t n_items) {
  int64_t want = (n_items + BLOCK - 1) / BLOCK;
  int64_t cap = 8192;
  return (int)std::min(std::max<int64_t>(want, 1), cap);
}

#define GSL(i, n, stride) for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < (n); i += (stride))
#define STRIDE ((int64_t)gridDim.x * blockDim.x)

// ---------------------------------------------------------------------------
// Elementwise: y = a*x + b*y  (general building block)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void axpby_kernel(
    float* __restrict__ y, const float* __restrict__ x, float a, float b, int64_t n) {
  int64_t n4 = n / 4;
  float4* y4 = reinterpret_cast<float4*>(y);
  const float4* x4 = reinterpret_cast<const float4*>(x);
  GSL(k, n4, STRIDE) {
    float4 xv = x4[k], yv = y4[k];
    yv.x = fmaf(a, xv.x, b * yv.x);
    yv.y = fmaf(a, xv.y, b * yv.y);
    yv.z = fmaf(a, xv.z, b * yv.z);
    yv.w = fmaf(a, xv.w, b * yv.w);
    y4[k] = yv;
  }
  GSL(k, n - n4 * 4, STRIDE) { int64_t i = n4 * 4 + k; y[i] = fmaf(a, x[i], b * y[i]); }
}

extern "C" void launch_axpby(float* y, const float* x, float a, float b, int64_t n, hipStream_t s) {
  axpby_kernel<<<grid_1d(n / 4 + 1), BLOCK, 0, s>>>(y, x, a, b, n);
}

// ---------------------------------------------------------------------------
// Fused proximal SGD step (K3/K4): FedProx / Ditto / MR-MTL inner step.
//   g' = g + weight_decay*p + mu*(p - w0)
//   if momentum>0: m = momentum*m + g';  u = nesterov ? g' + momentum*m : m
//   p -= lr*u
// Replaces reference WeightDriftLoss backward + separate optimizer.step()
// (fl4health/losses/weight_drift_loss.py + torch SGD): one HBM pass.
// w0 may be null (plain SGD). mbuf may be null (no momentum).
// ---------------------------------------------------------------------------
template <typename GT>
__device__ __forceinline__ float load_grad(const GT* g, int64_t i);
template <>
__device__ __forceinline__ float load_grad<float>(const float* g, int64_t i) { return g[i]; }
struct Bf16Tag { unsigned short v; };
template <>
__device__ __forceinline__ float load_grad<Bf16Tag>(const Bf16Tag* g, int64_t i) {
  unsigned int bits = (unsigned int)g[i].v << 16;
  return __uint_as_float(bits);
}

// fp32 master params; gradients may be fp32 OR bf16 (persistent bf16 weight
// mirrors produce bf16 grads — the cast up happens inline here, fused).
// Optional mirror output: after the update, p is re-cast bf16 into `mirror`
// so the next forward reads fresh compute weights (one pass, no extra kernel).
template <typename GT>
__global__ __launch_bounds__(BLOCK) void prox_sgd_kernel(
    float* __restrict__ p, const GT* __restrict__ g, const float* __restrict__ w0,
    float* __restrict__ mbuf, unsigned short* __restrict__ mirror, float lr, float mu,
    const float* __restrict__ mu_dev, float momentum, float weight_decay, int nesterov,
    int64_t n) {
  // mu may live in device memory (mu_dev) so a hipGraph-captured step sees
  // per-round server-adapted mu without re-capture
  if (mu_dev != nullptr) mu = mu_dev[0];
  GSL(i, n, STRIDE) {
    float gi = load_grad<GT>(g, i);
    float pi = p[i];
    if (weight_decay != 0.0f) gi = fmaf(weight_decay, pi, gi);
    if (w0 != nullptr) gi = fmaf(mu, pi - w0[i], gi);
    float u = gi;
    if (mbuf != nullptr) {
      float m = fmaf(momentum, mbuf[i], gi);
      mbuf[i] = m;
      u = nesterov ? fmaf(momentum, m, gi) : m;
    }
    float pnew = fmaf(-lr, u, pi);
    p[i] = pnew;
    if (mirror != nullptr) {
      // round-to-nearest-even f32 -> bf16
      unsigned int bits = __float_as_uint(pnew);
      unsigned int rounded = bits + 0x7FFFu + ((bits >> 16) & 1u);
      mirror[i] = (unsigned short)(rounded >> 16);
    }
  }
}

extern "C" void launch_prox_sgd(float* p, const void* g, int grad_is_bf16, const float* w0,
                                float* mbuf, unsigned short* mirror, float lr, float mu,
                                const float* mu_dev, float momentum, float weight_decay,
                                int nesterov, int64_t n, hipStream_t s) {
  if (grad_is_bf16) {
    prox_sgd_kernel<Bf16Tag><<<grid_1d(n), BLOCK, 0, s>>>(
        p, (const Bf16Tag*)g, w0, mbuf, mirror, lr, mu, mu_dev, momentum, weight_decay, nesterov, n);
  } else {
    prox_sgd_kernel<float><<<grid_1d(n), BLOCK, 0, s>>>(
        p, (const float*)g, w0, mbuf, mirror, lr, mu, mu_dev, momentum, weight_decay, nesterov, n);
  }
}

// ---------------------------------------------------------------------------
// SCAFFOLD variate-corrected SGD step (K3):
//   p -= lr * (g + c - ci)        [reference clients/scaffold_client.py:175-197]
// Fuses the modify_grad pass and the SGD update into one kernel.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void scaffold_sgd_kernel(
    float* __restrict__ p, const float* __restrict__ g, const float* __restrict__ c,
    const float* __restrict__ ci, float lr, float weight_decay, int64_t n) {
  GSL(i, n, STRIDE) {
    float gi = g[i] + c[i] - ci[i];
    float pi = p[i];
    if (weight_decay != 0.0f) gi = fmaf(weight_decay, pi, gi);
    p[i] = fmaf(-lr, gi, pi);
  }
}

extern "C" void launch_scaffold_sgd(float* p, const float* g, const float* c, const float* ci,
                                    float lr, float weight_decay, int64_t n, hipStream_t s) {
  scaffold_sgd_kernel<<<grid_1d(n), BLOCK, 0, s>>>(p, g, c, ci, lr, weight_decay, n);
}

// ---------------------------------------------------------------------------
// SCAFFOLD client control-variate update (K2):
//   ci_new = ci - c + (x_start - y_end) / (K * lr);  delta_ci = ci_new - ci (written in place over dci)
// (reference clients/scaffold_client.py:137-173)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void scaffold_variate_kernel(
    float* __restrict__ ci, float* __restrict__ dci, const float* __restrict__ c,
    const float* __restrict__ x_start, const float* __restrict__ y_end,
    float inv_klr, int64_t n) {
  GSL(i, n, STRIDE) {
    float ci_old = ci[i];
    float ci_new = ci_old - c[i] + (x_start[i] - y_end[i]) * inv_klr;
    ci[i] = ci_new;
    dci[i] = ci_new - ci_old;
  }
}

extern "C" void launch_scaffold_variate(float* ci, float* dci, const float* c,
                                        const float* x_start, const float* y_end,
                                        float inv_klr, int64_t n, hipStream_t s) {
  scaffold_variate_kernel<<<grid_1d(n), BLOCK, 0, s>>>(ci, dci, c, x_start, y_end, inv_klr, n);
}

// ---------------------------------------------------------------------------
// Server optimizer step (K13): FedOpt family + Flash, fused single pass.
// delta = aggregated client update direction (x_agg - x  or weighted delta sum).
//   kind 0 FedAvgM : m = b1*m + delta;                        u = m
//   kind 1 FedAdam : m = b1*m+(1-b1)*d; v = b2*v+(1-b2)*d^2;  u = m/(sqrt(v)+tau)
//   kind 2 FedYogi : m = ...; v = v-(1-b2)*sign(v-d^2)*d^2;   u = m/(sqrt(v)+tau)
//   kind 3 FedAdagrad: v = v + d^2;                           u = delta/(sqrt(v)+tau)  (m unused)
//   kind 4 Flash   : m=b1*m+(1-b1)*d; vp=v; v=b2*v+(1-b2)*d^2;
//                    b3m = |vp| / (|d^2-v| + |vp|)   (per-element, 0 when both 0)
//                    dt=b3m*dt+(1-b3m)*(d^2-v);               u = m/(sqrt(v)-dt+tau)
//     (reference fl4health/strategies/flash.py:125-170: beta_3 is a per-element
//      matrix derived from |v_prev| and |delta^2 - v_new|)
//   x += lr * u
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void server_opt_kernel(
    float* __restrict__ x, const float* __restrict__ delta, float* __restrict__ m,
    float* __restrict__ v, float* __restrict__ dt, int kind, float b1, float b2, float b3,
    float lr, float tau, int64_t n) {
  GSL(i, n, STRIDE) {
    float d = delta[i];
    float u;
    if (kind == 0) {
      float mi = fmaf(b1, m[i], d);
      m[i] = mi;
      u = mi;
    } else if (kind == 3) {
      float vi = fmaf(d, d, v[i]);
      v[i] = vi;
      u = d / (sqrtf(vi) + tau);
    } else {
      float mi = fmaf(b1, m[i], (1.0f - b1) * d);
      m[i] = mi;
      float d2 = d * d;
      float vi = v[i];
      if (kind == 1) {
        vi = fmaf(b2, vi, (1.0f - b2) * d2);
        v[i] = vi;
        u = mi / (sqrtf(vi) + tau);
      } else if (kind == 2) {
        float sgn = (vi > d2) ? 1.0f : ((vi < d2) ? -1.0f : 0.0f);
        vi = vi - (1.0f - b2) * sgn * d2;
        v[i] = vi;
        u = mi / (sqrtf(vi) + tau);
      } else {  // Flash
        float vprev = vi;
        vi = fmaf(b2, vi, (1.0f - b2) * d2);
        v[i] = vi;
        float diff = d2 - vi;
        float denom = fabsf(diff) + fabsf(vprev);
        float b3m = (denom > 0.0f) ? fabsf(vprev) / denom : 0.0f;
        float dti = fmaf(b3m, dt[i], (1.0f - b3m) * diff);
        dt[i] = dti;
        u = mi / (sqrtf(vi) - dti + tau);
      }
    }
    x[i] = fmaf(lr, u, x[i]);
  }
}

extern "C" void launch_server_opt(float* x, const float* delta, float* m, float* v, float* dt,
                                  int kind, float b1, float b2, float b3, float lr, float tau,
                                  int64_t n, hipStream_t s) {
  server_opt_kernel<<<grid_1d(n), BLOCK, 0, s>>>(x, delta, m, v, dt, kind, b1, b2, b3, lr, tau, n);
}

// ---------------------------------------------------------------------------
// Deterministic two-stage reductions (K5/K16 + drift-loss value K4).
// Stage 1: fixed NPART partial sums (fp64 accumulate); stage 2: one block.
// mode 0: sum x[i]^2        (sq norm)
// mode 1: sum (x-y)^2       (weight drift / clipping delta norm)
// mode 2: sum x*y           (dot, APFL alpha update K16)
// mode 3: sum x             (plain sum)
// ---------------------------------------------------------------------------
#define NPART 1024

__global__ __launch_bounds__(BLOCK) void reduce_partial_kernel(
    const float* __restrict__ x, const float* __restrict__ y, double* __restrict__ partial,
    int mode, int64_t n) {
  __shared__ double sm[BLOCK];
  double acc = 0.0;
  GSL(i, n, STRIDE) {
    float xi = x[i];
    if (mode == 0) acc += (double)xi * xi;
    else if (mode == 1) { float d = xi - y[i]; acc += (double)d * d; }
    else if (mode == 2) acc += (double)xi * y[i];
    else acc += (double)xi;
  }
  sm[threadIdx.x] = acc;
  __syncthreads();
  for (int off = BLOCK / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sm[threadIdx.x] += sm[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) partial[blockIdx.x] = sm[0];
}

__global__ __launch_bounds__(BLOCK) void reduce_final_kernel(
    const double* __restrict__ partial, double* __restrict__ out, int np) {
  __shared__ double sm[BLOCK];
  double acc = 0.0;
  for (int i = threadIdx.x; i < np; i += BLOCK) acc += partial[i];
  sm[threadIdx.x] = acc;
  __syncthreads();
  for (int off = BLOCK / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sm[threadIdx.x] += sm[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) out[0] = sm[0];
}

extern "C" void launch_reduce(const float* x, const float* y, double* partial, double* out,
                              int mode, int64_t n, hipStream_t s) {
  int np = std::min((int64_t)NPART, std::max<int64_t>((n + BLOCK - 1) / BLOCK, 1));
  reduce_partial_kernel<<<np, BLOCK, 0, s>>>(x, y, partial, mode, n);
  reduce_final_kernel<<<1, BLOCK, 0, s>>>(partial, out, np);
}

// ---------------------------------------------------------------------------
// Flat-clip of a weight delta (K5, client-level DP clipping):
//   delta = w - w0 ; scale = min(1, C/||delta||) ; out = w0 + scale*delta
// Norm comes from launch_reduce(mode 1); this kernel applies the scale given
// the already-computed norm (device scalar) without a host round-trip.
// (reference clients/clipping_client.py:71-111)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void clip_delta_kernel(
    float* __restrict__ out, const float* __restrict__ w, const float* __restrict__ w0,
    const double* __restrict__ sqnorm, float clip_bound, float* __restrict__ clip_bit,
    int64_t n) {
  float nrm = (float)sqrt(sqnorm[0]);
  float scale = (nrm > clip_bound && nrm > 0.0f) ? clip_bound / nrm : 1.0f;
  if (blockIdx.x == 0 && threadIdx.x == 0 && clip_bit != nullptr)
    clip_bit[0] = (nrm <= clip_bound) ? 1.0f : 0.0f;
  GSL(i, n, STRIDE) {
    float d = w[i] - w0[i];
    out[i] = scale * d;  // clipped DELTA (server aggregates deltas)
  }
}

extern "C" void launch_clip_delta(float* out, const float* w, const float* w0,
                                  const double* sqnorm, float clip_bound, float* clip_bit,
                                  int64_t n, hipStream_t s) {
  clip_delta_kernel<<<grid_1d(n), BLOCK, 0, s>>>(out, w, w0, sqnorm, clip_bound, clip_bit, n);
}

// ---------------------------------------------------------------------------
// Gaussian noise add (K6): x = a*x + sigma*N(0,1), Philox counter-based.
// (reference strategies/noisy_aggregate.py — np.random.normal there)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void gaussian_noise_kernel(
    float* __restrict__ x, float a, float sigma, uint64_t seed, uint64_t offset, int64_t n) {
  // each philox block covers 4 elements; use 2 box-muller pairs
  int64_t nblk = (n + 3) / 4;
  GSL(b, nblk, STRIDE) {
    Philox4 r = philox4x32(seed, offset + (uint64_t)b);
    float z0, z1, z2, z3;
    box_muller(r.x, r.y, &z0, &z1);
    box_muller(r.z, r.w, &z2, &z3);
    float zs[4] = {z0, z1, z2, z3};
    int64_t base = b * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int64_t i = base + j;
      if (i < n) x[i] = fmaf(sigma, zs[j], a * x[i]);
    }
  }
}

extern "C" void launch_gaussian_noise(float* x, float a, float sigma, uint64_t seed,
                                      uint64_t offset, int64_t n, hipStream_t s) {
  gaussian_noise_kernel<<<grid_1d((n + 3) / 4), BLOCK, 0, s>>>(x, a, sigma, seed, offset, n);
}

// ---------------------------------------------------------------------------
// FedPM mask sampling (K10): m = Bernoulli(sigmoid(score)); weff = m * w.
// Straight-through backward handled in Python (identity to scores).
// (reference fl4health/utils/functions.py:10-42 + model_bases/masked_layers/*)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void bernoulli_mask_kernel(
    const float* __restrict__ scores, const float* __restrict__ w, float* __restrict__ mask,
    float* __restrict__ weff, uint64_t seed, uint64_t offset, int apply_sigmoid, int64_t n) {
  int64_t nblk = (n + 3) / 4;
  GSL(b, nblk, STRIDE) {
    Philox4 r = philox4x32(seed, offset + (uint64_t)b);
    uint32_t us[4] = {r.x, r.y, r.z, r.w};
    int64_t base = b * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int64_t i = base + j;
      if (i < n) {
        float p = scores[i];
        if (apply_sigmoid) p = 1.0f / (1.0f + expf(-p));
        float u = u32_to_uniform(us[j]);
        float m = (u <= p) ? 1.0f : 0.0f;
        mask[i] = m;
        if (weff != nullptr) weff[i] = m * w[i];
      }
    }
  }
}

extern "C" void launch_bernoulli_mask(const float* scores, const float* w, float* mask,
                                      float* weff, uint64_t seed, uint64_t offset,
                                      int apply_sigmoid, int64_t n, hipStream_t s) {
  bernoulli_mask_kernel<<<grid_1d((n + 3) / 4), BLOCK, 0, s>>>(scores, w, mask, weff, seed,
                                                               offset, apply_sigmoid, n);
}

// ---------------------------------------------------------------------------
// DP-SGD per-sample ops (K7). Layout: per-sample grads for one layer flattened
// to [B, D] row-major (torch hooks produce this).
//  1) per_sample_sqnorm: out[b] += sum_d g[b,d]^2   (accumulated across layers)
//  2) clip_scaled_rowsum: out[d] += sum_b coef[b] * g[b,d]
// Both deterministic: fixed split of D into chunks; per (b,chunk) partial via
// one block row-slice; rowsum loops b inside the block (no atomics).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void per_sample_sqnorm_kernel(
    const float* __restrict__ g, float* __restrict__ out, int64_t B, int64_t D) {
  // grid.y = B ; grid.x covers D
  int64_t b = blockIdx.y;
  __shared__ float sm[BLOCK];
  float acc = 0.0f;
  const float* row = g + b * D;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < D;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = row[i];
    acc = fmaf(v, v, acc);
  }
  sm[threadIdx.x] = acc;
  __syncthreads();
  for (int off = BLOCK / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sm[threadIdx.x] += sm[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(&out[b], sm[0]);
}

extern "C" void launch_per_sample_sqnorm(const float* g, float* out, int64_t B, int64_t D,
                                         hipStream_t s) {
  int gx = (int)std::min<int64_t>((D + BLOCK - 1) / BLOCK, 256);
  dim3 grid(gx, (unsigned)B, 1);
  per_sample_sqnorm_kernel<<<grid, BLOCK, 0, s>>>(g, out, B, D);
}

// coef[b] = min(1, C / sqrt(sqnorm[b] + eps)) computed on the fly from the
// accumulated per-sample sq norms; out[d] += sum_b coef[b]*g[b,d].
__global__ __launch_bounds__(BLOCK) void clip_rowsum_kernel(
    const float* __restrict__ g, const float* __restrict__ sqnorms, float* __restrict__ out,
    float clip_bound, int64_t B, int64_t D) {
  GSL(d, D, STRIDE) {
    float acc = 0.0f;
    for (int64_t b = 0; b < B; ++b) {
      float nrm = sqrtf(sqnorms[b]) + 1e-6f;
      float coef = (nrm > clip_bound) ? clip_bound / nrm : 1.0f;
      acc = fmaf(coef, g[b * D + d], acc);
    }
    out[d] += acc;
  }
}

extern "C" void launch_clip_rowsum(const float* g, const float* sqnorms, float* out,
                                   float clip_bound, int64_t B, int64_t D, hipStream_t s) {
  clip_rowsum_kernel<<<grid_1d(D), BLOCK, 0, s>>>(g, sqnorms, out, clip_bound, B, D);
}

// Fused clip_rowsum + Gaussian noise (K7 epilogue): the clipped per-sample
// sum and the DP noise land in one pass instead of re-reading the whole grad
// in a second kernel. Philox addressing matches gaussian_noise_kernel
// (4 elements per counter block) so the noise stream is identical to the
// unfused path for the same (seed, offset).
__global__ __launch_bounds__(BLOCK) void clip_rowsum_noise_kernel(
    const float* __restrict__ g, const float* __restrict__ sqnorms, float* __restrict__ out,
    float clip_bound, float sigma, uint64_t seed, uint64_t offset, int64_t B, int64_t D) {
  int64_t nblk = (D + 3) / 4;
  GSL(q, nblk, STRIDE) {
    Philox4 r = philox4x32(seed, offset + (uint64_t)q);
    float z0, z1, z2, z3;
    box_muller(r.x, r.y, &z0, &z1);
    box_muller(r.z, r.w, &z2, &z3);
    float zs[4] = {z0, z1, z2, z3};
    int64_t base = q * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int64_t d = base + j;
      if (d >= D) continue;
      float acc = 0.0f;
      for (int64_t b = 0; b < B; ++b) {
        float nrm = sqrtf(sqnorms[b]) + 1e-6f;
        float coef = (nrm > clip_bound) ? clip_bound / nrm : 1.0f;
        acc = fmaf(coef, g[b * D + d], acc);
      }
      out[d] += acc + sigma * zs[j];
    }
  }
}

extern "C" void launch_clip_rowsum_noise(const float* g, const float* sqnorms, float* out,
                                         float clip_bound, float sigma, uint64_t seed,
                                         uint64_t offset, int64_t B, int64_t D, hipStream_t s) {
  clip_rowsum_noise_kernel<<<grid_1d((D + 3) / 4), BLOCK, 0, s>>>(g, sqnorms, out, clip_bound,
                                                                  sigma, seed, offset, B, D);
}

// ---------------------------------------------------------------------------
// Streaming confusion counts (K14): per-class TP/FP/FN/TN from argmax preds.
// out layout: int64 [C, 4] = (tp, fp, fn, tn). Single pass, LDS-staged
// per-block counters, one global atomic flush per block per class.
// (reference fl4health/metrics/efficient_metrics_base.py:308-375)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void confusion_kernel(
    const int64_t* __restrict__ preds, const int64_t* __restrict__ targets,
    unsigned long long* __restrict__ out, int C, int64_t n) {
  extern __shared__ unsigned int lcnt[];  // [C][2]: pred count, target count? need tp separately
  // layout: lcnt[c*3+0]=tp, [c*3+1]=pred_c (fp+tp), [c*3+2]=tgt_c (fn+tp)
  for (int i = threadIdx.x; i < C * 3; i += BLOCK) lcnt[i] = 0u;
  __syncthreads();
  int64_t total = 0;
  GSL(i, n, STRIDE) {
    int p = (int)preds[i];
    int t = (int)targets[i];
    if (p == t) atomicAdd(&lcnt[p * 3 + 0], 1u);
    atomicAdd(&lcnt[p * 3 + 1], 1u);
    atomicAdd(&lcnt[t * 3 + 2], 1u);
    total++;
  }
  __syncthreads();
  __shared__ unsigned long long blk_n;
  if (threadIdx.x == 0) blk_n = 0ull;
  __syncthreads();
  atomicAdd(&blk_n, (unsigned long long)total);
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += BLOCK) {
    unsigned long long tp = lcnt[c * 3 + 0];
    unsigned long long pc = lcnt[c * 3 + 1];
    unsigned long long tc = lcnt[c * 3 + 2];
    unsigned long long fp = pc - tp;
    unsigned long long fn = tc - tp;
    unsigned long long tn = blk_n - tp - fp - fn;
    if (tp) atomicAdd(&out[c * 4 + 0], tp);
    if (fp) atomicAdd(&out[c * 4 + 1], fp);
    if (fn) atomicAdd(&out[c * 4 + 2], fn);
    if (tn) atomicAdd(&out[c * 4 + 3], tn);
  }
}

extern "C" void launch_confusion(const int64_t* preds, const int64_t* targets,
                                 unsigned long long* out, int C, int64_t n, hipStream_t s) {
  size_t shmem = (size_t)C * 3 * sizeof(unsigned int);
  confusion_kernel<<<grid_1d(n), BLOCK, shmem, s>>>(preds, targets, out, C, n);
}

// ---------------------------------------------------------------------------
// Weighted multi-buffer average epilogue (K1 epilogue in gather mode):
// out = sum_k w[k] * bufs[k]  over up to 8 stacked flat buffers [K, n].
// Used by gather-path strategies on rank 0 (deterministic fixed order).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void weighted_sum_rows_kernel(
    const float* __restrict__ stack, const float* __restrict__ w, float* __restrict__ out,
    int K, int64_t n) {
  GSL(i, n, STRIDE) {
    float acc = 0.0f;
    for (int k = 0; k < K; ++k) acc = fmaf(w[k], stack[(int64_t)k * n + i], acc);
    out[i] = acc;
  }
}

extern "C" void launch_weighted_sum_rows(const float* stack, const float* w, float* out, int K,
                                         int64_t n, hipStream_t s) {
  weighted_sum_rows_kernel<<<grid_1d(n), BLOCK, 0, s>>>(stack, w, out, K, n);
}

// ---------------------------------------------------------------------------
// K11: deterministic stream compaction for sparse COO packing (reference
// parameter_packer.py:94-142 + sparse_coo_parameter_exchanger.py:18 do this
// with host-side nonzero/scatter). Two passes around ONE torch cumsum:
//   1) per-thread-chunk hit counts (each thread owns a contiguous chunk, so
//      the final ordering is row-major ascending — identical to nonzero())
//   2) each thread writes its chunk's selected values + per-dim indices at
//      its exclusive offset.
// ---------------------------------------------------------------------------
#define COMPACT_CHUNK 64

__global__ __launch_bounds__(BLOCK) void coo_count_kernel(
    const float* __restrict__ score, float threshold, int64_t n, int64_t n_chunks,
    int32_t* __restrict__ counts) {
  GSL(c, n_chunks, STRIDE) {
    int64_t lo = c * COMPACT_CHUNK;
    int64_t hi = min(lo + COMPACT_CHUNK, n);
    int cnt = 0;
    for (int64_t i = lo; i < hi; ++i) cnt += (score[i] >= threshold) ? 1 : 0;
    counts[c] = cnt;
  }
}

__global__ __launch_bounds__(BLOCK) void coo_write_kernel(
    const float* __restrict__ values_in, const float* __restrict__ score, float threshold,
    const int32_t* __restrict__ offsets,  // exclusive per-chunk offsets
    float* __restrict__ values_out, int64_t* __restrict__ indices_out,  // [ndim, nnz]
    const int64_t* __restrict__ dims, int ndim, int64_t n, int64_t n_chunks, int64_t nnz) {
  GSL(c, n_chunks, STRIDE) {
    int64_t lo = c * COMPACT_CHUNK;
    int64_t hi = min(lo + COMPACT_CHUNK, n);
    int64_t out = offsets[c];
    for (int64_t i = lo; i < hi; ++i) {
      if (score[i] < threshold) continue;
      values_out[out] = values_in[i];
      int64_t rem = i;
      for (int d = ndim - 1; d >= 0; --d) {
        indices_out[(int64_t)d * nnz + out] = rem % dims[d];
        rem /= dims[d];
      }
      ++out;
    }
  }
}

extern "C" void launch_coo_count(const float* score, float threshold, int64_t n,
                                 int64_t n_chunks, int32_t* counts, hipStream_t s) {
  coo_count_kernel<<<grid_1d(n_chunks), BLOCK, 0, s>>>(score, threshold, n, n_chunks, counts);
}

extern "C" void launch_coo_write(const float* values_in, const float* score, float threshold,
                                 const int32_t* offsets, float* values_out, int64_t* indices_out,
                                 const int64_t* dims, int ndim, int64_t n, int64_t n_chunks,
                                 int64_t nnz, hipStream_t s) {
  coo_write_kernel<<<grid_1d(n_chunks), BLOCK, 0, s>>>(values_in, score, threshold, offsets,
                                                       values_out, indices_out, dims, ndim, n,
                                                       n_chunks, nnz);
}
