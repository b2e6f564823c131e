// Fused MOON/NT-Xent-style contrastive loss (reference
// losses/contrastive_loss.py:6-167).
//
// One workgroup per sample: cosine similarities of z_b against the positive
// and K negative partners (block reductions over D), softmax-CE with label 0,
// AND the input gradient dz in the same launch. The partners are frozen
// snapshots in MOON (no grad), so the whole loss is ONE kernel instead of
// eager's normalize/expand/cossim/cat/CE chain (~10 launches + intermediate
// [K, B, D] materialization).
#include <hip/hip_runtime.h>
#include <stdint.h>

#define CON_THREADS 256
#define CON_MAX_PARTNERS 17  // 1 positive + up to 16 negatives

__global__ __launch_bounds__(CON_THREADS) void moon_contrastive_kernel(
    const float* __restrict__ z,    // [B, D]
    const float* __restrict__ pos,  // [B, D]
    const float* __restrict__ neg,  // [K, B, D]
    int K, int B, int64_t D, float inv_tau,
    float* __restrict__ loss_out,   // [B]
    float* __restrict__ dz) {       // [B, D]
  __shared__ float s_red[CON_THREADS];
  __shared__ float s_probs[CON_MAX_PARTNERS];
  __shared__ float s_sims[CON_MAX_PARTNERS];
  __shared__ float s_invn[CON_MAX_PARTNERS];  // 1/(|z||w_j|)
  __shared__ float s_invnz2;

  int b = blockIdx.x;
  int tid = threadIdx.x;
  const float* zb = z + (int64_t)b * D;

  // ||z||^2
  float acc = 0.f;
  for (int64_t d = tid; d < D; d += CON_THREADS) acc += zb[d] * zb[d];
  s_red[tid] = acc;
  __syncthreads();
  for (int s = CON_THREADS / 2; s > 0; s >>= 1) {
    if (tid < s) s_red[tid] += s_red[tid + s];
    __syncthreads();
  }
  float nz2 = s_red[0] + 1e-12f;
  if (tid == 0) s_invnz2 = 1.0f / nz2;
  __syncthreads();

  // per-partner dot + norm
  for (int j = 0; j <= K; ++j) {
    const float* wj = (j == 0) ? pos + (int64_t)b * D : neg + (((int64_t)(j - 1) * B + b) * D);
    float dot = 0.f, nw = 0.f;
    for (int64_t d = tid; d < D; d += CON_THREADS) {
      float w = wj[d];
      dot += zb[d] * w;
      nw += w * w;
    }
    s_red[tid] = dot;
    __syncthreads();
    for (int s = CON_THREADS / 2; s > 0; s >>= 1) {
      if (tid < s) s_red[tid] += s_red[tid + s];
      __syncthreads();
    }
    float dot_tot = s_red[0];
    s_red[tid] = nw;
    __syncthreads();
    for (int s = CON_THREADS / 2; s > 0; s >>= 1) {
      if (tid < s) s_red[tid] += s_red[tid + s];
      __syncthreads();
    }
    if (tid == 0) {
      float nw_tot = s_red[0] + 1e-12f;
      float inv_nznw = rsqrtf(nz2 * nw_tot);
      s_invn[j] = inv_nznw;
      s_sims[j] = dot_tot * inv_nznw;
    }
    __syncthreads();
  }

  // softmax-CE over [pos | negs] / tau with label 0
  if (tid == 0) {
    float mx = -1e30f;
    for (int j = 0; j <= K; ++j) mx = fmaxf(mx, s_sims[j] * inv_tau);
    float denom = 0.f;
    for (int j = 0; j <= K; ++j) {
      s_probs[j] = __expf(s_sims[j] * inv_tau - mx);
      denom += s_probs[j];
    }
    float inv_denom = 1.0f / denom;
    for (int j = 0; j <= K; ++j) s_probs[j] *= inv_denom;
    loss_out[b] = -__logf(fmaxf(s_probs[0], 1e-30f));
  }
  __syncthreads();

  // dz_b = (1/(B tau)) sum_j (p_j - [j==0]) (w_j/(|z||w_j|) - sim_j z_b/|z|^2)
  float inv_b_tau = inv_tau / (float)B;
  float invnz2 = s_invnz2;
  for (int64_t d = tid; d < D; d += CON_THREADS) {
    float zv = zb[d];
    float g = 0.f;
    for (int j = 0; j <= K; ++j) {
      const float* wj = (j == 0) ? pos + (int64_t)b * D : neg + (((int64_t)(j - 1) * B + b) * D);
      float coeff = s_probs[j] - (j == 0 ? 1.0f : 0.0f);
      g += coeff * (wj[d] * s_invn[j] - s_sims[j] * zv * invnz2);
    }
    dz[(int64_t)b * D + d] = g * inv_b_tau;
  }
}

extern "C" void launch_moon_contrastive(const float* z, const float* pos, const float* neg,
                                        int K, int B, int64_t D, float inv_tau, float* loss_out,
                                        float* dz, hipStream_t s) {
  moon_contrastive_kernel<<<dim3(B), CON_THREADS, 0, s>>>(z, pos, neg, K, B, D, inv_tau,
                                                          loss_out, dz);
}
