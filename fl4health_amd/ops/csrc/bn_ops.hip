// Hand-written CDNA4 NHWC BatchNorm (training fwd + bwd).
//
// Why: rocprof on the flagship bench (profiles/) shows MIOpen's
// BatchNorm*Spatial* kernels + their SubTensorOp side-kernels + autocast's
// bf16<->fp32 casts are ~40% of the ResNet-18 step while moving a few hundred
// MB — an order of magnitude off the HBM roofline. This implementation:
//   - NHWC (channels-last) layout: channel index is innermost, so per-channel
//     reductions are column sums of an [R, C] matrix with perfectly coalesced
//     rows (R = N*H*W).
//   - bf16 OR fp32 I/O with fp32 statistics math (no autocast cast kernels:
//     the op consumes conv's bf16 output directly).
//   - deterministic two-stage reductions: fixed G row-groups -> finalize.
//   - 3 data passes fwd (reduce, normalize incl. fused write of x_hat-free
//     form), 5 passes bwd — vs MIOpen's many-kernel pipeline.
// All launches are stream-ordered and allocation-free (workspaces come from
// the caller) => hipGraph-capture safe.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>
#include <algorithm>

#define BNBLOCK 128
// Row-stripes per reduce block: each block LDS-combines BNRW per-thread
// stripe sums before writing ONE partial row, shrinking the partial array
// (and the finalize kernel's read volume) by BNRW vs one-stripe blocks.
#define BNRW 8

template <typename T>
__device__ __forceinline__ float ld(const T* p);
template <>
__device__ __forceinline__ float ld<float>(const float* p) { return *p; }
template <>
__device__ __forceinline__ float ld<__hip_bfloat16>(const __hip_bfloat16* p) {
  return __bfloat162float(*p);
}

template <typename T>
__device__ __forceinline__ void st(T* p, float v);
template <>
__device__ __forceinline__ void st<float>(float* p, float v) { *p = v; }
template <>
__device__ __forceinline__ void st<__hip_bfloat16>(__hip_bfloat16* p, float v) {
  *p = __float2bfloat16(v);
}

// ---------------------------------------------------------------------------
// fwd pass 1: per-channel partial sums. Deterministic fixed-stripe split:
// thread (by, ty) owns rows r == by*BNRW+ty (mod Gb*BNRW); the block combines
// its BNRW stripes through LDS and writes one row of partial [2, Gb, C].
// ---------------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(BNBLOCK * BNRW) void bn_fwd_reduce_kernel(
    const T* __restrict__ x, float* __restrict__ partial, int64_t R, int C, int Gb) {
  __shared__ float sm[BNRW][2][BNBLOCK];
  int c = blockIdx.x * BNBLOCK + threadIdx.x;
  int S = Gb * BNRW;
  int stripe = blockIdx.y * BNRW + threadIdx.y;
  float s = 0.0f, ss = 0.0f;
  if (c < C) {
    for (int64_t r = stripe; r < R; r += S) {
      float v = ld<T>(x + r * C + c);
      s += v;
      ss = fmaf(v, v, ss);
    }
  }
  sm[threadIdx.y][0][threadIdx.x] = s;
  sm[threadIdx.y][1][threadIdx.x] = ss;
  __syncthreads();
  if (threadIdx.y == 0 && c < C) {
#pragma unroll
    for (int j = 1; j < BNRW; ++j) {
      s += sm[j][0][threadIdx.x];
      ss += sm[j][1][threadIdx.x];
    }
    partial[(int64_t)blockIdx.y * C + c] = s;
    partial[(int64_t)(Gb + blockIdx.y) * C + c] = ss;
  }
}

// fwd pass 2: finalize mean/invstd, update running stats.
// One block PER CHANNEL; 256 threads tree-reduce the G partials (G can be
// thousands — a serial per-channel loop here was the original bottleneck).
#define BNFIN 256
__global__ __launch_bounds__(BNFIN) void bn_fwd_finalize_kernel(
    const float* __restrict__ partial, float* __restrict__ mean, float* __restrict__ invstd,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    int64_t* __restrict__ num_batches_tracked, float momentum, float eps,
    int64_t R, int C, int G) {
  // buffer-update fold: the counter bump that was a separate 4 us
  // CUDAFunctorOnSelf_add<long> launch per BN layer per step
  if (blockIdx.x == 0 && threadIdx.x == 0 && num_batches_tracked != nullptr)
    ++*num_batches_tracked;
  __shared__ float sm_s[BNFIN];
  __shared__ float sm_ss[BNFIN];
  int c = blockIdx.x;
  float s = 0.0f, ss = 0.0f;
  for (int g = threadIdx.x; g < G; g += BNFIN) {
    s += partial[(int64_t)g * C + c];
    ss += partial[(int64_t)(G + g) * C + c];
  }
  sm_s[threadIdx.x] = s;
  sm_ss[threadIdx.x] = ss;
  __syncthreads();
  for (int off = BNFIN / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      sm_s[threadIdx.x] += sm_s[threadIdx.x + off];
      sm_ss[threadIdx.x] += sm_ss[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x != 0) return;
  s = sm_s[0];
  ss = sm_ss[0];
  float m = s / (float)R;
  float var = fmaxf(ss / (float)R - m * m, 0.0f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    running_mean[c] = (1.0f - momentum) * running_mean[c] + momentum * m;
    float unbiased = (R > 1) ? var * (float)R / (float)(R - 1) : var;
    running_var[c] = (1.0f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// fwd pass 3: y = (x - mean) * invstd * gamma + beta, optionally fused ReLU
template <typename T>
__global__ __launch_bounds__(BNBLOCK) void bn_fwd_norm_kernel(
    const T* __restrict__ x, T* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, const T* __restrict__ res, int fuse_relu, int64_t R, int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float m = mean[c];
  float is = invstd[c];
  float g = gamma != nullptr ? gamma[c] : 1.0f;
  float b = beta != nullptr ? beta[c] : 0.0f;
  float scale = is * g;
  float shift = b - m * scale;
  for (int64_t r = blockIdx.y; r < R; r += gridDim.y) {
    float v = fmaf(ld<T>(x + r * C + c), scale, shift);
    if (res != nullptr) v += ld<T>(res + r * C + c);  // fused residual add
    if (fuse_relu) v = fmaxf(v, 0.0f);
    st<T>(y + r * C + c, v);
  }
}

// ---------------------------------------------------------------------------
// bwd pass 1: per-channel partials of (sum dy, sum dy * x_hat)
// ---------------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(BNBLOCK * BNRW) void bn_bwd_reduce_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, float* __restrict__ partial,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta, const T* __restrict__ res,
    T* __restrict__ dres, int fuse_relu, int64_t R, int C, int Gb) {
  __shared__ float sm[BNRW][2][BNBLOCK];
  int c = blockIdx.x * BNBLOCK + threadIdx.x;
  int S = Gb * BNRW;
  int stripe = blockIdx.y * BNRW + threadIdx.y;
  float sdy = 0.0f, sdyx = 0.0f;
  if (c < C) {
    float m = mean[c];
    float is = invstd[c];
    // with fused ReLU, dy must be masked where the pre-activation was <= 0:
    // recompute the sign from (x_hat * gamma + beta) instead of saving y
    float gm = gamma != nullptr ? gamma[c] : 1.0f;
    float bt = beta != nullptr ? beta[c] : 0.0f;
    for (int64_t r = stripe; r < R; r += S) {
      float gy = ld<T>(dy + r * C + c);
      float xh = (ld<T>(x + r * C + c) - m) * is;
      float pre = fmaf(xh, gm, bt);
      if (res != nullptr) pre += ld<T>(res + r * C + c);
      if (fuse_relu && pre <= 0.0f) gy = 0.0f;
      // the residual branch receives exactly the relu-masked dy
      if (dres != nullptr) st<T>(dres + r * C + c, gy);
      sdy += gy;
      sdyx = fmaf(gy, xh, sdyx);
    }
  }
  sm[threadIdx.y][0][threadIdx.x] = sdy;
  sm[threadIdx.y][1][threadIdx.x] = sdyx;
  __syncthreads();
  if (threadIdx.y == 0 && c < C) {
#pragma unroll
    for (int j = 1; j < BNRW; ++j) {
      sdy += sm[j][0][threadIdx.x];
      sdyx += sm[j][1][threadIdx.x];
    }
    partial[(int64_t)blockIdx.y * C + c] = sdy;
    partial[(int64_t)(Gb + blockIdx.y) * C + c] = sdyx;
  }
}

__global__ __launch_bounds__(BNFIN) void bn_bwd_finalize_kernel(
    const float* __restrict__ partial, float* __restrict__ sum_dy, float* __restrict__ sum_dy_xhat,
    float* __restrict__ dgamma, float* __restrict__ dbeta, int C, int G) {
  __shared__ float sm_s[BNFIN];
  __shared__ float sm_ss[BNFIN];
  int c = blockIdx.x;
  float sdy = 0.0f, sdyx = 0.0f;
  for (int g = threadIdx.x; g < G; g += BNFIN) {
    sdy += partial[(int64_t)g * C + c];
    sdyx += partial[(int64_t)(G + g) * C + c];
  }
  sm_s[threadIdx.x] = sdy;
  sm_ss[threadIdx.x] = sdyx;
  __syncthreads();
  for (int off = BNFIN / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      sm_s[threadIdx.x] += sm_s[threadIdx.x + off];
      sm_ss[threadIdx.x] += sm_ss[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x != 0) return;
  sdy = sm_s[0];
  sdyx = sm_ss[0];
  sum_dy[c] = sdy;
  sum_dy_xhat[c] = sdyx;
  if (dgamma != nullptr) dgamma[c] = sdyx;
  if (dbeta != nullptr) dbeta[c] = sdy;
}

// bwd pass 2: dx = gamma*invstd * (dy - sum_dy/R - x_hat * sum_dy_xhat/R)
template <typename T>
__global__ __launch_bounds__(BNBLOCK) void bn_bwd_dx_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, T* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ sum_dy, const float* __restrict__ sum_dy_xhat,
    const T* __restrict__ res, int fuse_relu, int64_t R, int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float m = mean[c];
  float is = invstd[c];
  float g = gamma != nullptr ? gamma[c] : 1.0f;
  float bt = beta != nullptr ? beta[c] : 0.0f;
  float k = g * is;
  float mean_dy = sum_dy[c] / (float)R;
  float mean_dy_xhat = sum_dy_xhat[c] / (float)R;
  for (int64_t r = blockIdx.y; r < R; r += gridDim.y) {
    float gy = ld<T>(dy + r * C + c);
    float xh = (ld<T>(x + r * C + c) - m) * is;
    float pre = fmaf(xh, g, bt);
    if (res != nullptr) pre += ld<T>(res + r * C + c);
    if (fuse_relu && pre <= 0.0f) gy = 0.0f;
    st<T>(dx + r * C + c, k * (gy - mean_dy - xh * mean_dy_xhat));
  }
}

// ---------------------------------------------------------------------------
// launchers (dtype: 0 = fp32, 1 = bf16); G fixed for determinism
// ---------------------------------------------------------------------------
static inline void bn_dims(int C, int64_t R, int G, dim3* grid, dim3* block) {
  block->x = BNBLOCK;
  block->y = 1;
  block->z = 1;
  grid->x = (C + BNBLOCK - 1) / BNBLOCK;
  grid->y = G;
  grid->z = 1;
}

// G here is the partial-group count Gb; the reduce kernels internally stripe
// rows S = Gb*BNRW ways and LDS-combine, so thread count matches the 1D form.
static inline void bn_reduce_dims(int C, int G, dim3* grid, dim3* block) {
  block->x = BNBLOCK;
  block->y = BNRW;
  block->z = 1;
  grid->x = (C + BNBLOCK - 1) / BNBLOCK;
  grid->y = G;
  grid->z = 1;
}

extern "C" void launch_bn_fwd(const void* x, void* y, float* partial, float* mean, float* invstd,
                              const float* gamma, const float* beta, float* running_mean,
                              float* running_var, int64_t* num_batches_tracked, float momentum,
                              float eps, int64_t R, int C,
                              int G, int dtype, int fuse_relu, const void* res, hipStream_t s) {
  dim3 grid, block, rgrid, rblock;
  bn_dims(C, R, G * BNRW, &grid, &block);
  bn_reduce_dims(C, G, &rgrid, &rblock);
  if (dtype == 1) {
    bn_fwd_reduce_kernel<__hip_bfloat16><<<rgrid, rblock, 0, s>>>(
        (const __hip_bfloat16*)x, partial, R, C, G);
  } else {
    bn_fwd_reduce_kernel<float><<<rgrid, rblock, 0, s>>>((const float*)x, partial, R, C, G);
  }
  bn_fwd_finalize_kernel<<<dim3(C, 1, 1), dim3(BNFIN, 1, 1), 0, s>>>(
      partial, mean, invstd, running_mean, running_var, num_batches_tracked, momentum, eps, R, C,
      G);
  // normalize: G*BNRW-deep row grid (bandwidth-bound)
  if (dtype == 1) {
    bn_fwd_norm_kernel<__hip_bfloat16><<<grid, block, 0, s>>>(
        (const __hip_bfloat16*)x, (__hip_bfloat16*)y, mean, invstd, gamma, beta,
        (const __hip_bfloat16*)res, fuse_relu, R, C);
  } else {
    bn_fwd_norm_kernel<float><<<grid, block, 0, s>>>((const float*)x, (float*)y, mean, invstd,
                                                     gamma, beta, (const float*)res, fuse_relu,
                                                     R, C);
  }
}

extern "C" void launch_bn_bwd(const void* x, const void* dy, void* dx, float* partial,
                              const float* mean, const float* invstd, const float* gamma,
                              const float* beta, float* sum_dy, float* sum_dy_xhat, float* dgamma,
                              float* dbeta, int64_t R, int C, int G, int dtype, int fuse_relu,
                              const void* res, void* dres, hipStream_t s) {
  dim3 grid, block, rgrid, rblock;
  bn_dims(C, R, G * BNRW, &grid, &block);
  bn_reduce_dims(C, G, &rgrid, &rblock);
  if (dtype == 1) {
    bn_bwd_reduce_kernel<__hip_bfloat16><<<rgrid, rblock, 0, s>>>(
        (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy, partial, mean, invstd, gamma, beta,
        (const __hip_bfloat16*)res, (__hip_bfloat16*)dres, fuse_relu, R, C, G);
  } else {
    bn_bwd_reduce_kernel<float><<<rgrid, rblock, 0, s>>>((const float*)x, (const float*)dy, partial,
                                                       mean, invstd, gamma, beta,
                                                       (const float*)res, (float*)dres, fuse_relu,
                                                       R, C, G);
  }
  bn_bwd_finalize_kernel<<<dim3(C, 1, 1), dim3(BNFIN, 1, 1), 0, s>>>(
      partial, sum_dy, sum_dy_xhat, dgamma, dbeta, C, G);
  if (dtype == 1) {
    bn_bwd_dx_kernel<__hip_bfloat16><<<grid, block, 0, s>>>(
        (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy, (__hip_bfloat16*)dx, mean, invstd,
        gamma, beta, sum_dy, sum_dy_xhat, (const __hip_bfloat16*)res, fuse_relu, R, C);
  } else {
    bn_bwd_dx_kernel<float><<<grid, block, 0, s>>>((const float*)x, (const float*)dy, (float*)dx,
                                                   mean, invstd, gamma, beta, sum_dy, sum_dy_xhat,
                                                   (const float*)res, fuse_relu, R, C);
  }
}
