// Direct 3x3 stride-1 pad-1 NHWC bf16 convolution forward on MFMA (gfx950).
//
// Round-2 lever from docs/BACKLOG.md: MIOpen's igemm reaches ~12% of the bf16
// MFMA peak on CIFAR-scale spatial sizes; this hand-written implicit-GEMM
// direct conv already beats it on the mid-depth ResNet shapes
// (profiles/kernels_summary.md).
//
// Mapping (one workgroup = 4 waves = 256 threads):
//   output tile: M = SB*BH*W positions (SB packed samples x BH image rows,
//   sized to 128) x KB = 64 output channels; each wave owns a 32 x 64 slice
//   = 2(M) x 4(N) mfma_f32_16x16x32_bf16 tiles.
//   reduction: loop c-chunks of 64; per chunk stage input halo tiles
//   (BH+2) x (W+2) x 64 per sample and weights 9 x 64 x KB into LDS, then
//   9 taps x 2 mfma-K steps accumulate. Input reuse across the 9 taps comes
//   from the halo tile in LDS (the win over im2col: no patch
//   materialization). All staging is 16-byte vectorized.
//   When C fits one chunk (C <= 64), weights are staged ONCE and the block
//   grid-strides over M-tiles, amortizing the weight stage.
// Fragment layouts validated by tools/csrc/mfma_probe.hip on hardware:
//   A: row = lane&15, k = (lane>>4)*8 + j ; B: col = lane&15, same k;
//   C/D: col = lane&15, row = (lane>>4)*4 + reg.
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <algorithm>

using bf16 = __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define CONV_KB 64      // output channels per block
#define CONV_CB 64      // reduction channels per LDS stage
#define CONV_THREADS 256

// LDS: input halo tiles [SB][(BH+2)][(W+2)][CB] + weights [9][KB][CB] (both
// bf16). Tile extent is SB*(BH+2)*(W+2) <= 304 positions across supported
// shapes (W=32: 6*34=204; W=4, SB=8: 8*6*6=288).
#define CONV_TILE_POS 304
// one static LDS object (≈112 KB of the 160 KB/CU; single workgroup per CU)
__shared__ __bf16 s_conv[CONV_TILE_POS * CONV_CB + 9 * CONV_CB * CONV_KB];

__device__ inline void stage_weights(const bf16* __restrict__ w, __bf16* __restrict__ s_w,
                                     int C, int K, int c0, int cb, int kb0, int tid) {
  // global w is [9][C][K] (k contiguous); LDS is k-major [9][KB][CB] so B
  // fragments (fixed out-channel, 8 consecutive c) are one 16-byte read.
  for (int idx = tid; idx < 9 * CONV_CB * (CONV_KB / 8); idx += CONV_THREADS) {
    int kv = idx % (CONV_KB / 8);
    int c = (idx / (CONV_KB / 8)) % CONV_CB;
    int tap = idx / ((CONV_KB / 8) * CONV_CB);
    int k = kv * 8;
    bf16x8 v = bf16x8{};
    if (c < cb && kb0 + k + 8 <= K)
      v = *reinterpret_cast<const bf16x8*>(&w[((int64_t)tap * C + c0 + c) * K + kb0 + k]);
    else if (c < cb)
      for (int j = 0; j < 8; ++j)
        v[j] = (kb0 + k + j < K) ? w[((int64_t)tap * C + c0 + c) * K + kb0 + k + j] : (bf16)0.0f;
#pragma unroll
    for (int j = 0; j < 8; ++j) s_w[(tap * CONV_KB + k + j) * CONV_CB + c] = v[j];
  }
}

__global__ __launch_bounds__(CONV_THREADS) void conv3x3_fwd_kernel(
    const bf16* __restrict__ x,   // [N, H, W, C]
    const bf16* __restrict__ w,   // [9, C, K] taps-major, prepacked
    const float* __restrict__ bias,  // [K] or nullptr
    bf16* __restrict__ y,         // [N, H, W, K]
    int Nn, int H, int W, int C, int K, int BH, int SB, int n_tiles, int h_groups) {
  __bf16* s_in = s_conv;                          // [SB*(BH+2)*(W+2)][CB]
  __bf16* s_w = s_conv + CONV_TILE_POS * CONV_CB; // [9][KB][CB]

  int kb0 = blockIdx.z * CONV_KB;
  int tile_w = W + 2;
  int tid = threadIdx.x;
  int wave = tid >> 6;
  int lane = tid & 63;
  bool persistent_w = (C <= CONV_CB);
  if (persistent_w) {
    stage_weights(w, s_w, C, K, 0, C, kb0, tid);
    __syncthreads();
  }

  for (int tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
    int hg = tile % h_groups;
    int n0 = (tile / h_groups) * SB;
    int sb = min(SB, Nn - n0);
    int h0 = hg * BH;
    int bh = min(BH, H - h0);
    int pps = bh * W;             // positions per sample-section
    int m_count = sb * pps;

    f32x4 acc[2][4];
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int q = 0; q < 4; ++q) acc[t][q] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int c0 = 0; c0 < C; c0 += CONV_CB) {
      int cb = min(CONV_CB, C - c0);
      // ---- stage input halo tiles, 16 B per thread (8 channels) ----
      for (int idx = tid; idx < sb * (bh + 2) * tile_w * (CONV_CB / 8); idx += CONV_THREADS) {
        int cv = idx % (CONV_CB / 8);        // 8-channel vector index
        int pos = idx / (CONV_CB / 8);
        int col = pos % tile_w;              // 0..W+1 -> image col col-1
        int row = (pos / tile_w) % (bh + 2); // 0..bh+1 -> image row h0+row-1
        int s = pos / (tile_w * (bh + 2));
        int ih = h0 + row - 1, iw = col - 1;
        bf16x8 v = bf16x8{};
        int c = cv * 8;
        if (ih >= 0 && ih < H && iw >= 0 && iw < W && c + 8 <= cb)
          v = *reinterpret_cast<const bf16x8*>(
              &x[(((int64_t)(n0 + s) * H + ih) * W + iw) * C + c0 + c]);
        else if (ih >= 0 && ih < H && iw >= 0 && iw < W)
          for (int j = 0; j < 8; ++j)
            v[j] = (c + j < cb) ? x[(((int64_t)(n0 + s) * H + ih) * W + iw) * C + c0 + c + j]
                                : (bf16)0.0f;
        *reinterpret_cast<bf16x8*>(&s_in[pos * CONV_CB + c]) = v;
      }
      if (!persistent_w) stage_weights(w, s_w, C, K, c0, cb, kb0, tid);
      __syncthreads();

      // ---- 9 taps x (CB/32) mfma-K steps ----
#pragma unroll
      for (int tap = 0; tap < 9; ++tap) {
        int dy = tap / 3, dx = tap % 3;
#pragma unroll
        for (int ck = 0; ck < CONV_CB / 32; ++ck) {
          int kbase = ck * 32 + (lane >> 4) * 8;
          bf16x8 afrag[2];
#pragma unroll
          for (int t = 0; t < 2; ++t) {
            int m = wave * 32 + t * 16 + (lane & 15);
            int s = m / pps, rem = m % pps;
            int hh = rem / W, ww = rem % W;
            const __bf16* src =
                &s_in[((s * (bh + 2) + hh + dy) * tile_w + (ww + dx)) * CONV_CB + kbase];
            // channel-contiguous: one 16-byte LDS read per fragment
            bf16x8 a = *reinterpret_cast<const bf16x8*>(src);
            if (m >= m_count) a = bf16x8{};
            afrag[t] = a;
          }
#pragma unroll
          for (int q = 0; q < 4; ++q) {
            const __bf16* bw = &s_w[(tap * CONV_KB + q * 16 + (lane & 15)) * CONV_CB + kbase];
            bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(bw);
#pragma unroll
            for (int t = 0; t < 2; ++t)
              acc[t][q] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[t], bfrag, acc[t][q], 0, 0, 0);
          }
        }
      }
      // covers both the next c-chunk's s_in overwrite and, on the last
      // chunk, the next TILE's s_in overwrite (epilogue touches no LDS)
      __syncthreads();
    }

    // ---- epilogue: fp32 acc (+bias) -> bf16 NHWC ----
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int q = 0; q < 4; ++q)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int m = wave * 32 + t * 16 + (lane >> 4) * 4 + r;
          int k = q * 16 + (lane & 15);
          if (m < m_count && kb0 + k < K) {
            int s = m / pps, rem = m % pps;
            int hh = rem / W, ww = rem % W;
            float v = acc[t][q][r];
            if (bias != nullptr) v += bias[kb0 + k];
            y[(((int64_t)(n0 + s) * H + h0 + hh) * W + ww) * (int64_t)K + kb0 + k] = (bf16)v;
          }
        }
  }
}

extern "C" void launch_conv3x3_fwd(const void* x, const void* w, const float* bias, void* y,
                                   int Nn, int H, int W, int C, int K, hipStream_t s) {
  int BH = std::min(std::max(128 / W, 1), H);
  int SB = std::max(128 / (H * W), 1);  // pack small images, several per block
  SB = std::min(SB, Nn);
  int h_groups = (H + BH - 1) / BH;
  int n_tiles = ((Nn + SB - 1) / SB) * h_groups;
  int kz = (K + CONV_KB - 1) / CONV_KB;
  // one tile per block: capping the grid to ~CU count and grid-striding was
  // measured SLOWER on 32x32 (0.075 -> 0.089 ms) — serializing tiles costs
  // more than the per-block weight stage saves at these sizes. The
  // grid-stride machinery stays for the future glds-pipelined variant.
  dim3 grid(n_tiles, 1, kz);
  conv3x3_fwd_kernel<<<grid, CONV_THREADS, 0, s>>>(
      (const bf16*)x, (const bf16*)w, bias, (bf16*)y, Nn, H, W, C, K, BH, SB, n_tiles, h_groups);
}
