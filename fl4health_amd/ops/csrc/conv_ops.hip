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
#include <cstdlib>

using bf16 = __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// LDS bank-conflict swizzle (guide st_16x32): the halo/weight images are
// 128-B rows (64 bf16 channels); a ds_read_b128 whose 16 lanes stride rows
// 8-way-conflicts on the 64x4B banks. XOR channel bit 4 with row bit 2 to
// spread the lane group over 4 bank slots. Applied BOTH sides: reads XOR the
// LDS address; staging XORs (a) the ds_write channel for the plain path and
// (b) the per-lane GLOBAL fetch address for glds (the glds LDS destination is
// lane-linear and cannot be swizzled — guide rule 21).
__device__ inline int conv_swz(int row) { return ((row >> 2) & 1) << 4; }

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
                                     int C, int K, int c0, int cb, int kb0, int tid,
                                     int nthreads = CONV_THREADS) {
  // global w is [9][C][K] (k contiguous); LDS is k-major [9][KB][CB] so B
  // fragments (fixed out-channel, 8 consecutive c) are one 16-byte read.
  for (int idx = tid; idx < 9 * CONV_CB * (CONV_KB / 8); idx += nthreads) {
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
    for (int j = 0; j < 8; ++j) {
      int r = tap * CONV_KB + k + j;
      s_w[r * CONV_CB + (c ^ conv_swz(r))] = v[j];
    }
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
        *reinterpret_cast<bf16x8*>(&s_in[pos * CONV_CB + (c ^ conv_swz(pos))]) = v;
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
            int apos = (s * (bh + 2) + hh + dy) * tile_w + (ww + dx);
            // channel-contiguous: one 16-byte LDS read per fragment
            bf16x8 a = *reinterpret_cast<const bf16x8*>(
                &s_in[apos * CONV_CB + (kbase ^ conv_swz(apos))]);
            if (m >= m_count) a = bf16x8{};
            afrag[t] = a;
          }
#pragma unroll
          for (int q = 0; q < 4; ++q) {
            int br = tap * CONV_KB + q * 16 + (lane & 15);
            bf16x8 bfrag =
                *reinterpret_cast<const bf16x8*>(&s_w[br * CONV_CB + (kbase ^ conv_swz(br))]);
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

// ---------------------------------------------------------------------------
// Variant B: persistent-weight + glds double-buffered input (C == CONV_CB).
//
// The round-1 kernel at 32x32 C64 runs 1 block/CU with a serial
// stage->sync->compute chain (0.55x MIOpen). Here the weights are staged once
// per block, the grid is capped near the CU count, the block grid-strides over
// M-tiles, and the NEXT tile's input halo is fetched with async
// global_load_lds (16 B, lane-linear LDS image) WHILE the current tile's MFMAs
// run; the single __syncthreads() per tile drains the DMA (guide §5: 2-buffer
// glds + vmcnt(0) ties the best register pipeline in this 1-block/CU regime).
// Out-of-range / halo lanes redirect their source address to a zero page —
// the LDS destination of glds is wave-uniform base + lane*16 and cannot be
// masked per-lane.
// ---------------------------------------------------------------------------
#define CONV_GLDS_CHUNK_CAP 1792   // 4-wave variant: 16 B chunks/buffer (32x32 M=128: 1632)
#define CONV_GLDS_CHUNK_CAP8 2752  // 8-wave variant (32x32 M=256: 10*34*8 = 2720)
// NOTE both caps are gated on round_up(chunks, 64): glds bases are 64-lane
// aligned, so the ragged last group writes up to round_up(chunks, 64) - 1;
// an exact-sized buffer lets it spill into the neighbouring buffer / the
// persistent weights (the round-1 8-wave corruption bug).
__device__ __align__(16) __bf16 conv_zero16[8] = {};

__device__ inline void issue_glds_input(const bf16* __restrict__ x, __bf16* __restrict__ dst,
                                        int n0, int h0, int sb, int bh, int H, int W, int C,
                                        int c0, int tile_w, int tid, int nthreads) {
  int wave = tid >> 6, lane = tid & 63;
  int cpv = CONV_CB / 8;  // LDS image stride is CONV_CB channels per position
  int total_chunks = sb * (bh + 2) * tile_w * cpv;
  for (int base = wave * 64; base < total_chunks; base += nthreads) {
    int idx = base + lane;
    const bf16* src = conv_zero16;
    if (idx < total_chunks) {
      int cv = idx % cpv;
      int pos = idx / cpv;
      int col = pos % tile_w;
      int row = (pos / tile_w) % (bh + 2);
      int sidx = pos / (tile_w * (bh + 2));
      int ih = h0 + row - 1, iw = col - 1;
      int cf = (cv * 8) ^ conv_swz(pos);  // LDS slot cv*8 holds logical channel cf
      if (ih >= 0 && ih < H && iw >= 0 && iw < W && cf < C - c0)
        src = &x[(((int64_t)(n0 + sidx) * H + ih) * W + iw) * C + c0 + cf];
    }
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)src,
                                     (__attribute__((address_space(3))) void*)(dst + (size_t)base * 8),
                                     16, 0, 0);
  }
}

template <int WAVES>
__global__ __launch_bounds__(WAVES * 64) void conv3x3_fwd_glds_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w, const float* __restrict__ bias,
    bf16* __restrict__ y, int Nn, int H, int W, int C, int K, int BH, int SB, int n_tiles,
    int h_groups) {
  constexpr int NT = WAVES * 64;
  constexpr int CAP = (WAVES == 8) ? CONV_GLDS_CHUNK_CAP8 : CONV_GLDS_CHUNK_CAP;
  // 4-wave: 131 KB; 8-wave: 157 KB of the 160 KB LDS (1 WG/CU either way —
  // the 8-wave variant exists to put 2 waves on each SIMD for latency hiding)
  __shared__ __bf16 s_mem[2 * CAP * 8 + 9 * CONV_CB * CONV_KB];
  __bf16* s_w = s_mem + 2 * CAP * 8;
  int kb0 = blockIdx.z * CONV_KB;
  int tile_w = W + 2;
  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;

  stage_weights(w, s_w, C, K, 0, C, kb0, tid, NT);  // persistent: staged ONCE
  float breg[4];
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    int k = kb0 + q * 16 + (lane & 15);
    breg[q] = (bias != nullptr && k < K) ? bias[k] : 0.0f;
  }

  auto tile_geom = [&](int tile, int& n0, int& h0, int& sb, int& bh) {
    int hg = tile % h_groups;
    n0 = (tile / h_groups) * SB;
    sb = min(SB, Nn - n0);
    h0 = hg * BH;
    bh = min(BH, H - h0);
  };

  int tile = blockIdx.x;
  if (tile < n_tiles) {
    int n0, h0, sb, bh;
    tile_geom(tile, n0, h0, sb, bh);
    issue_glds_input(x, s_mem, n0, h0, sb, bh, H, W, C, 0, tile_w, tid, NT);
  }
  __syncthreads();  // drains prologue glds (vmcnt 0) + weight ds_writes
  int cur = 0;

  for (; tile < n_tiles; tile += gridDim.x) {
    int n0, h0, sb, bh;
    tile_geom(tile, n0, h0, sb, bh);
    int pps = bh * W;
    int m_count = sb * pps;
    int nxt = tile + gridDim.x;
    if (nxt < n_tiles) {
      int nn0, nh0, nsb, nbh;
      tile_geom(nxt, nn0, nh0, nsb, nbh);
      issue_glds_input(x, s_mem + (1 - cur) * (CAP * 8), nn0, nh0, nsb, nbh, H, W, C, 0,
                       tile_w, tid, NT);
    }
    const __bf16* s_in = s_mem + cur * (CAP * 8);

    f32x4 acc[2][4];
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int q = 0; q < 4; ++q) acc[t][q] = f32x4{0.f, 0.f, 0.f, 0.f};

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
          int sidx = m / pps, rem = m % pps;
          int hh = rem / W, ww = rem % W;
          int apos = (sidx * (bh + 2) + hh + dy) * tile_w + (ww + dx);
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &s_in[apos * CONV_CB + (kbase ^ conv_swz(apos))]);
          if (m >= m_count) a = bf16x8{};
          afrag[t] = a;
        }
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          int br = tap * CONV_KB + q * 16 + (lane & 15);
          bf16x8 bfrag =
              *reinterpret_cast<const bf16x8*>(&s_w[br * CONV_CB + (kbase ^ conv_swz(br))]);
#pragma unroll
          for (int t = 0; t < 2; ++t)
            acc[t][q] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[t], bfrag, acc[t][q], 0, 0, 0);
        }
      }
    }

#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int q = 0; q < 4; ++q)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int m = wave * 32 + t * 16 + (lane >> 4) * 4 + r;
          int k = q * 16 + (lane & 15);
          if (m < m_count && kb0 + k < K) {
            int sidx = m / pps, rem = m % pps;
            int hh = rem / W, ww = rem % W;
            y[(((int64_t)(n0 + sidx) * H + h0 + hh) * W + ww) * (int64_t)K + kb0 + k] =
                (bf16)(acc[t][q][r] + breg[q]);
          }
        }
    __syncthreads();  // all waves done reading s_in; prefetch glds drained
    cur ^= 1;
  }
}

// ---------------------------------------------------------------------------
// Variant C: KB = 32, BOTH operands glds double-buffered, weights prepacked
// host-side into the exact LDS image ([kz][cchunk][tap][kk 0..31][cc 0..63]
// contiguous slabs) so weight staging is a pure lane-linear DMA — no address
// math, no ds_write pass, and it overlaps the previous chunk's MFMAs. One
// barrier per c-chunk. Built for the deep small-spatial shapes (4x4 C512:
// round-1 kernel filled only 128 of 256 CUs; KB = 32 doubles kz and fills the
// chip without split-C partials).
// ---------------------------------------------------------------------------
#define CONVC_KB 32
#define CONVC_CHUNK_CAP 2304  // max input chunks (4x4 SB=8: 288 pos * 8)
#define CONVC_WCHUNKS (9 * CONVC_KB * CONV_CB / 8)  // 2304 weight chunks/slab
__shared__ __bf16 s_convc[2 * CONVC_CHUNK_CAP * 8 + 2 * CONVC_WCHUNKS * 8];

__device__ inline void issue_glds_slab(const bf16* __restrict__ src_base, __bf16* __restrict__ dst,
                                       int n_chunks, int tid) {
  int wave = tid >> 6, lane = tid & 63;
  for (int base = wave * 64; base < n_chunks; base += CONV_THREADS) {
    const bf16* src = src_base + (size_t)(base + lane) * 8;  // slab is contiguous; cap is exact
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)src,
                                     (__attribute__((address_space(3))) void*)(dst + (size_t)base * 8),
                                     16, 0, 0);
  }
}

__global__ __launch_bounds__(CONV_THREADS) void conv3x3_fwd_kb32_kernel(
    const bf16* __restrict__ x,
    const bf16* __restrict__ wimg,  // [K/32][C/64][9][32][64] contiguous slabs
    const float* __restrict__ bias, bf16* __restrict__ y, int Nn, int H, int W, int C, int K,
    int BH, int SB, int n_tiles, int h_groups) {

  int tile = blockIdx.x;
  int kzi = blockIdx.z;
  int kb0 = kzi * CONVC_KB;
  int tile_w = W + 2;
  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int n_cchunks = (C + CONV_CB - 1) / CONV_CB;

  int hg = tile % h_groups;
  int n0 = (tile / h_groups) * SB;
  int sb = min(SB, Nn - n0);
  int h0 = hg * BH;
  int bh = min(BH, H - h0);
  int pps = bh * W;
  int m_count = sb * pps;

  const bf16* wslab0 = wimg + ((size_t)kzi * n_cchunks) * (CONVC_WCHUNKS * 8);
  issue_glds_input(x, s_convc, n0, h0, sb, bh, H, W, C, 0, tile_w, tid, CONV_THREADS);
  issue_glds_slab(wslab0, s_convc + 2 * CONVC_CHUNK_CAP * 8, CONVC_WCHUNKS, tid);
  __syncthreads();

  f32x4 acc[2][2];
#pragma unroll
  for (int t = 0; t < 2; ++t)
#pragma unroll
    for (int q = 0; q < 2; ++q) acc[t][q] = f32x4{0.f, 0.f, 0.f, 0.f};

  int cur = 0;
  for (int ci = 0; ci < n_cchunks; ++ci) {
    if (ci + 1 < n_cchunks) {
      issue_glds_input(x, s_convc + (1 - cur) * (CONVC_CHUNK_CAP * 8), n0, h0, sb, bh, H, W, C,
                       (ci + 1) * CONV_CB, tile_w, tid, CONV_THREADS);
      issue_glds_slab(wslab0 + (size_t)(ci + 1) * (CONVC_WCHUNKS * 8),
                      s_convc + 2 * CONVC_CHUNK_CAP * 8 + (1 - cur) * (CONVC_WCHUNKS * 8),
                      CONVC_WCHUNKS, tid);
    }
    const __bf16* sin = s_convc + cur * (CONVC_CHUNK_CAP * 8);
    const __bf16* sw = s_convc + 2 * CONVC_CHUNK_CAP * 8 + cur * (CONVC_WCHUNKS * 8);
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
          int sidx = m / pps, rem = m % pps;
          int hh = rem / W, ww = rem % W;
          int apos = (sidx * (bh + 2) + hh + dy) * tile_w + (ww + dx);
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &sin[apos * CONV_CB + (kbase ^ conv_swz(apos))]);
          if (m >= m_count) a = bf16x8{};
          afrag[t] = a;
        }
#pragma unroll
        for (int q = 0; q < 2; ++q) {
          int br = tap * CONVC_KB + q * 16 + (lane & 15);
          bf16x8 bfrag =
              *reinterpret_cast<const bf16x8*>(&sw[br * CONV_CB + (kbase ^ conv_swz(br))]);
#pragma unroll
          for (int t = 0; t < 2; ++t)
            acc[t][q] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[t], bfrag, acc[t][q], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // drains next chunk's DMA; all waves done with cur
    cur ^= 1;
  }

#pragma unroll
  for (int t = 0; t < 2; ++t)
#pragma unroll
    for (int q = 0; q < 2; ++q)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int m = wave * 32 + t * 16 + (lane >> 4) * 4 + r;
        int k = q * 16 + (lane & 15);
        if (m < m_count && kb0 + k < K) {
          int sidx = m / pps, rem = m % pps;
          int hh = rem / W, ww = rem % W;
          float v = acc[t][q][r];
          if (bias != nullptr) v += bias[kb0 + k];
          y[(((int64_t)(n0 + sidx) * H + h0 + hh) * W + ww) * (int64_t)K + kb0 + k] = (bf16)v;
        }
      }
}

static inline void conv_tile_geom(int Nn, int H, int W, int M, int& BH, int& SB, int& h_groups,
                                  int& n_tiles) {
  BH = std::min(std::max(M / W, 1), H);
  SB = std::max(M / (H * W), 1);  // pack small images, several per block
  SB = std::min(SB, Nn);
  h_groups = (H + BH - 1) / BH;
  n_tiles = ((Nn + SB - 1) / SB) * h_groups;
}

extern "C" void launch_conv3x3_fwd(const void* x, const void* w, const float* bias, void* y,
                                   int Nn, int H, int W, int C, int K, hipStream_t s) {
  int kz = (K + CONV_KB - 1) / CONV_KB;
  static int cap = [] {
    const char* e = getenv("FL4_CONVB_GRID");
    return e ? atoi(e) : 256;
  }();
  static int force_waves = [] {
    const char* e = getenv("FL4_CONVB_WAVES");
    return e ? atoi(e) : -1;  // -1 auto, 0 disable variant B, 4/8 force
  }();
  if (C <= CONV_CB && C % 8 == 0 && force_waves != 0) {
    // Variant B: persistent weights + pipelined input DMA. Prefer the 8-wave
    // (M = 256) instantiation — 2 waves/SIMD hides ds_read/MFMA latency at
    // 1 block/CU — then the 4-wave one; fall through if the halo won't fit.
    int BH, SB, h_groups, n_tiles;
    conv_tile_geom(Nn, H, W, 256, BH, SB, h_groups, n_tiles);
    int chunks8 = SB * (BH + 2) * (W + 2) * (CONV_CB / 8);
    if (((chunks8 + 63) & ~63) <= CONV_GLDS_CHUNK_CAP8 && force_waves != 4) {
      dim3 grid(std::min(n_tiles, cap), 1, kz);
      conv3x3_fwd_glds_kernel<8><<<grid, 512, 0, s>>>(
          (const bf16*)x, (const bf16*)w, bias, (bf16*)y, Nn, H, W, C, K, BH, SB, n_tiles,
          h_groups);
      return;
    }
    conv_tile_geom(Nn, H, W, 128, BH, SB, h_groups, n_tiles);
    int chunks4 = SB * (BH + 2) * (W + 2) * (CONV_CB / 8);
    if (((chunks4 + 63) & ~63) <= CONV_GLDS_CHUNK_CAP) {
      dim3 grid(std::min(n_tiles, 2 * cap), 1, kz);
      conv3x3_fwd_glds_kernel<4><<<grid, CONV_THREADS, 0, s>>>(
          (const bf16*)x, (const bf16*)w, bias, (bf16*)y, Nn, H, W, C, K, BH, SB, n_tiles,
          h_groups);
      return;
    }
  }
  int BH, SB, h_groups, n_tiles;
  conv_tile_geom(Nn, H, W, 128, BH, SB, h_groups, n_tiles);
  dim3 grid(n_tiles, 1, kz);
  conv3x3_fwd_kernel<<<grid, CONV_THREADS, 0, s>>>(
      (const bf16*)x, (const bf16*)w, bias, (bf16*)y, Nn, H, W, C, K, BH, SB, n_tiles, h_groups);
}

// Variant C entry: caller supplies the LDS-image weight pack (see
// ops/conv.py pack_weight_kb32). Requires C % 64 == 0, K % 32 == 0.
extern "C" void launch_conv3x3_fwd_kb32(const void* x, const void* wimg, const float* bias,
                                        void* y, int Nn, int H, int W, int C, int K,
                                        hipStream_t s) {
  int BH, SB, h_groups, n_tiles;
  conv_tile_geom(Nn, H, W, 128, BH, SB, h_groups, n_tiles);
  int kz = K / CONVC_KB;
  dim3 grid(n_tiles, 1, kz);
  conv3x3_fwd_kb32_kernel<<<grid, CONV_THREADS, 0, s>>>(
      (const bf16*)x, (const bf16*)wimg, bias, (bf16*)y, Nn, H, W, C, K, BH, SB, n_tiles,
      h_groups);
}

// ---------------------------------------------------------------------------
// Fused weight pack for the KB=32 kernel: [K, C, 3, 3] bf16 (standard torch
// conv layout) -> [K/32][C/64][9][32][64] LDS-image slabs with the conv_swz
// bank swizzle baked in. One kernel replaces the ~6-op torch chain
// (permute/contiguous/arange/index_select/where) that measured ~0.5 ms/step
// of launch+small-tensor overhead in the flagship bench. mode 0: forward
// pack. mode 1: bwd-data pack — roles swapped (conv over gy has C_conv = K
// and K_conv = C) and taps rotated 180 degrees.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void pack_kb32_kernel(
    const bf16* __restrict__ w,  // [K, C, 3, 3]
    bf16* __restrict__ out,      // [Kc/32, Cc/64, 9, 32, 64] (conv-role dims)
    int K, int C, int mode) {
  int Kc = mode == 0 ? K : C;  // conv-role output channels
  int Cc = mode == 0 ? C : K;
  int64_t total = (int64_t)Kc * Cc * 9;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int cc_slot = idx % 64;
    int kk = (idx / 64) % 32;
    int tap = (idx / (64 * 32)) % 9;
    int ci = (idx / (64 * 32 * 9)) % (Cc / 64);
    int kz = idx / ((int64_t)64 * 32 * 9 * (Cc / 64));
    int cc = cc_slot ^ conv_swz(tap * CONVC_KB + kk);  // logical channel
    int kc = kz * 32 + kk;
    int c_conv = ci * 64 + cc;
    int dy = tap / 3, dx = tap % 3;
    bf16 v;
    if (mode == 0) {
      v = w[(((int64_t)kc * C + c_conv) * 3 + dy) * 3 + dx];
    } else {
      // bwd: conv-role (kc, c_conv) = (orig C, orig K); taps flipped
      v = w[(((int64_t)c_conv * C + kc) * 3 + (2 - dy)) * 3 + (2 - dx)];
    }
    out[idx] = v;
  }
}

extern "C" void launch_pack_kb32(const void* w, void* out, int K, int C, int mode,
                                 hipStream_t s) {
  int64_t total = (int64_t)K * C * 9;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  pack_kb32_kernel<<<blocks, 256, 0, s>>>((const bf16*)w, (bf16*)out, K, C, mode);
}

// ---------------------------------------------------------------------------
// Variant D: input-resident multi-kz KB=32 ("kzloop"). For mid shapes
// (16x16 C64/C128) the ENTIRE input halo fits LDS alongside two weight
// slabs, so one block stages its input ONCE and loops every K-block with
// the weight DMA pipelined — grid = n_tiles (fills at 256 tiles), no input
// restaging per kz (variant C re-reads input KZ times), prologue amortized
// over KZ*CC chunks.
// ---------------------------------------------------------------------------
#define CONVD_IN_CAP 1664      // 16-B chunks per input c-chunk (16x16: 1440, 8x8 SB2: 1600) + ragged-glds slack
#define CONVD_MAX_CC 2         // resident input c-chunks
__shared__ __bf16 s_convd[CONVD_MAX_CC * CONVD_IN_CAP * 8 + 2 * CONVC_WCHUNKS * 8];

template <int WAVES>
__global__ __launch_bounds__(WAVES * 64) void conv3x3_fwd_kzloop_kernel(
    const bf16* __restrict__ x,
    const bf16* __restrict__ wimg,  // [K/32][C/64][9][32][64] contiguous slabs
    const float* __restrict__ bias, bf16* __restrict__ y, int Nn, int H, int W, int C, int K,
    int BH, int SB, int n_tiles, int h_groups) {
  // WAVES == 4: each wave owns a 32(M) x 32(K) tile (acc[2][2]).
  // WAVES == 8: wave pairs split the K dim — wave (w&3) covers M rows, bit
  // w>>2 selects which 16-wide K half it computes (acc[2][1]) => 2 waves per
  // SIMD for ds_read/MFMA latency hiding at the same LDS footprint.
  constexpr int NT = WAVES * 64;
  constexpr int QS = (WAVES == 8) ? 1 : 2;  // q slots per wave
  int tile = blockIdx.x;
  int tile_w = W + 2;
  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int wave_m = (WAVES == 8) ? (wave & 3) : wave;
  int qbase = (WAVES == 8) ? (wave >> 2) : 0;
  int n_cchunks = (C + CONV_CB - 1) / CONV_CB;
  int kz = K / CONVC_KB;

  int hg = tile % h_groups;
  int n0 = (tile / h_groups) * SB;
  int sb = min(SB, Nn - n0);
  int h0 = hg * BH;
  int bh = min(BH, H - h0);
  int pps = bh * W;
  int m_count = sb * pps;

  __bf16* s_w0 = s_convd + CONVD_MAX_CC * CONVD_IN_CAP * 8;
  // stage ALL input c-chunks once + the first weight slab
  for (int ci = 0; ci < n_cchunks; ++ci)
    issue_glds_input(x, s_convd + (size_t)ci * (CONVD_IN_CAP * 8), n0, h0, sb, bh, H, W, C,
                     ci * CONV_CB, tile_w, tid, NT);
  issue_glds_slab(wimg, s_w0, CONVC_WCHUNKS, tid);
  __syncthreads();

  f32x4 acc[2][QS];
#pragma unroll
  for (int t = 0; t < 2; ++t)
#pragma unroll
    for (int q = 0; q < QS; ++q) acc[t][q] = f32x4{0.f, 0.f, 0.f, 0.f};

  int total = kz * n_cchunks;
  int cur = 0;
  for (int it = 0; it < total; ++it) {
    int kzi = it / n_cchunks;
    int ci = it % n_cchunks;
    if (it + 1 < total)
      issue_glds_slab(wimg + (size_t)(it + 1) * (CONVC_WCHUNKS * 8),
                      s_w0 + (size_t)(1 - cur) * (CONVC_WCHUNKS * 8), CONVC_WCHUNKS, tid);
    const __bf16* sin = s_convd + (size_t)ci * (CONVD_IN_CAP * 8);
    const __bf16* sw = s_w0 + (size_t)cur * (CONVC_WCHUNKS * 8);
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      int dy = tap / 3, dx = tap % 3;
#pragma unroll
      for (int ck = 0; ck < CONV_CB / 32; ++ck) {
        int kbase = ck * 32 + (lane >> 4) * 8;
        bf16x8 afrag[2];
#pragma unroll
        for (int t = 0; t < 2; ++t) {
          int m = wave_m * 32 + t * 16 + (lane & 15);
          int sidx = m / pps, rem = m % pps;
          int hh = rem / W, ww = rem % W;
          int apos = (sidx * (bh + 2) + hh + dy) * tile_w + (ww + dx);
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &sin[apos * CONV_CB + (kbase ^ conv_swz(apos))]);
          if (m >= m_count) a = bf16x8{};
          afrag[t] = a;
        }
#pragma unroll
        for (int q = 0; q < QS; ++q) {
          int br = tap * CONVC_KB + (qbase + q) * 16 + (lane & 15);
          bf16x8 bfrag =
              *reinterpret_cast<const bf16x8*>(&sw[br * CONV_CB + (kbase ^ conv_swz(br))]);
#pragma unroll
          for (int t = 0; t < 2; ++t)
            acc[t][q] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[t], bfrag, acc[t][q], 0, 0, 0);
        }
      }
    }
    if (ci == n_cchunks - 1) {
      int kb0 = kzi * CONVC_KB;
#pragma unroll
      for (int t = 0; t < 2; ++t)
#pragma unroll
        for (int q = 0; q < QS; ++q) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int m = wave_m * 32 + t * 16 + (lane >> 4) * 4 + r;
            int k = (qbase + q) * 16 + (lane & 15);
            if (m < m_count && kb0 + k < K) {
              int sidx = m / pps, rem = m % pps;
              int hh = rem / W, ww = rem % W;
              float v = acc[t][q][r];
              if (bias != nullptr) v += bias[kb0 + k];
              y[(((int64_t)(n0 + sidx) * H + h0 + hh) * W + ww) * (int64_t)K + kb0 + k] = (bf16)v;
            }
          }
          acc[t][q] = f32x4{0.f, 0.f, 0.f, 0.f};
        }
    }
    __syncthreads();
    cur ^= 1;
  }
}

extern "C" void launch_conv3x3_fwd_kzloop(const void* x, const void* wimg, const float* bias,
                                          void* y, int Nn, int H, int W, int C, int K,
                                          hipStream_t s) {
  int BH, SB, h_groups, n_tiles;
  conv_tile_geom(Nn, H, W, 128, BH, SB, h_groups, n_tiles);
  dim3 grid(n_tiles, 1, 1);
  static int waves = [] {
    const char* e = getenv("FL4_KZLOOP_WAVES");
    return e ? atoi(e) : 8;
  }();
  if (waves == 8) {
    conv3x3_fwd_kzloop_kernel<8><<<grid, 512, 0, s>>>(
        (const bf16*)x, (const bf16*)wimg, bias, (bf16*)y, Nn, H, W, C, K, BH, SB, n_tiles,
        h_groups);
  } else {
    conv3x3_fwd_kzloop_kernel<4><<<grid, CONV_THREADS, 0, s>>>(
        (const bf16*)x, (const bf16*)wimg, bias, (bf16*)y, Nn, H, W, C, K, BH, SB, n_tiles,
        h_groups);
  }
}

// ---------------------------------------------------------------------------
// conv3x3 backward-WEIGHTS (wrw): dW[tap][c][k] = sum_r x[r_shift, c] dy[r, k]
// v1 scope: NHWC bf16, stride1/pad1, C=K=64, W=32 (the ResNet layer-1 family
// - the shapes where MIOpen's wrw igemm runs ~4x off the HBM roofline because
// it cannot reuse one x tile across all 9 taps).
//
// Design: deterministic split-K over (n, h-group) tiles. Each WG stages one
// tile TRANSPOSED in LDS - x_t[c][(BH+2) x (W+2) halo patch] and
// dy_t[k][BH x W] - because the MFMA contraction dim here is the ROW dim:
// both operands feed __builtin_amdgcn_mfma_f32_16x16x32_bf16 as
// [channel][row] fragments, and one staged x patch serves all 9 taps (the
// tap shift is just a different base offset into the patch). 8 waves = 4
// m-groups (c) x 2 n-groups (k); acc = 2 n-tiles x 9 taps x f32x4 per wave.
// Each WG accumulates its tile range in registers and writes ONE fp32
// partial dW; a combine kernel reduces the fixed split count (deterministic).
// ---------------------------------------------------------------------------
#define WRW_SPLITS 256
// x is staged THREE times, one copy per horizontal tap shift, so every
// MFMA A-fragment read is a 16B-aligned ds_read_b128 (a single shifted copy
// forces odd-u16 offsets -> scalar LDS reads, measured 0.46x MIOpen).
// Pitches are multiples of 8 elements (alignment) whose dword stride is
// ≡ 4 (mod 32), spreading 16 reading lanes over 8 LDS banks (2-way).
// The transpose STORES additionally XOR-swizzle the 8-element block index
// by (c>>3)&7: the 8 lanes of a store group share sp but differ in c8, and
// their channel stride is ≡ 0 mod 32 banks — unswizzled they all hit one
// bank (8-way conflict on every staging store).
// Index math for both geometries is bit-verified by the host simulator that
// produced this layout (see commit history / tools/wrw_micro.py).
#define WRW_SPITCH 264
#define WRW_DPITCH 136
__device__ __forceinline__ size_t wrw_swz(int c, int sp) {
  return (size_t)c * WRW_SPITCH + (((sp >> 3) ^ ((c >> 3) & 7)) << 3) + (sp & 7);
}

// One 64-channel x 64-filter wrw slice: x[...,c0:c0+64] against
// dy[...,k0:k0+64] with physical strides XC/KC (C/K up to 128 run as 2x2
// sub-slices from the launcher — LDS cannot hold three shifted 128-channel
// copies). W_=32/BH_=4 covers ResNet layer 1, W_=16/BH_=8 layer 2; a
// 32-row MFMA k-chunk spans 32/W_ image lines and each lane's 8 contiguous
// rows stay within one line for W_ >= 8.
template <int W_, int BH_>
__global__ __launch_bounds__(512) void conv3x3_wrw_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ dy, bf16* __restrict__ partial,
    int Nn, int H, int XC, int c0, int KC, int k0, int n_tiles) {
  constexpr int LP = W_ + 8;  // staged line pitch (W_+2 halo cols, padded)
  __shared__ __bf16 s_xt[3 * 64 * WRW_SPITCH];
  __shared__ __bf16 s_dyt[64 * WRW_DPITCH];
  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int mg = wave & 3, ng = wave >> 2;
  int ln = lane & 15, km = lane >> 4;
  int h_groups = H / BH_;

  f32x4 acc[2][9];
#pragma unroll
  for (int t = 0; t < 2; ++t)
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) acc[t][tap] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int tile = blockIdx.x; tile < n_tiles; tile += WRW_SPLITS) {
    int n0 = tile / h_groups;
    int h0 = (tile % h_groups) * BH_;
    // stage x: lines h0-1 .. h0+BH_, w -1..W_; copy d holds x[w = col + d - 1]
    const int xtotal = (BH_ + 2) * (W_ + 2) * (64 / 8);
    for (int idx = tid; idx < xtotal; idx += 512) {
      int c8 = idx & 7;
      int sp = idx >> 3;
      int line = sp / (W_ + 2), wx = sp % (W_ + 2);
      int hh = h0 - 1 + line, ww = wx - 1;
      bf16x8 v = bf16x8{};
      if (hh >= 0 && hh < H && ww >= 0 && ww < W_)
        v = *reinterpret_cast<const bf16x8*>(
            &x[(((int64_t)n0 * H + hh) * W_ + ww) * XC + c0 + c8 * 8]);
#pragma unroll
      for (int d = 0; d < 3; ++d) {
        int col = ww + 1 - d;  // copy d: col w holds x[w + d - 1]
        if (col < 0 || col >= LP) continue;
        __bf16* dst = &s_xt[((size_t)d * 64) * WRW_SPITCH];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          dst[wrw_swz(c8 * 8 + j, line * LP + col)] = v[j];
      }
    }
    const int dtotal = BH_ * W_ * (64 / 8);
    for (int idx = tid; idx < dtotal; idx += 512) {
      int k8 = idx & 7;
      int sp = idx >> 3;
      int line = sp / W_, ww = sp % W_;
      bf16x8 v = *reinterpret_cast<const bf16x8*>(
          &dy[(((int64_t)n0 * H + h0 + line) * W_ + ww) * KC + k0 + k8 * 8]);
#pragma unroll
      for (int j = 0; j < 8; ++j) s_dyt[(size_t)(k8 * 8 + j) * WRW_DPITCH + sp] = v[j];
    }
    __syncthreads();
#pragma unroll
    for (int kc = 0; kc < BH_ * W_ / 32; ++kc) {
      int r0 = kc * 32 + km * 8;
      int line = r0 / W_, col = r0 % W_;
      bf16x8 bfrag[2];
#pragma unroll
      for (int t = 0; t < 2; ++t)
        bfrag[t] = *reinterpret_cast<const bf16x8*>(
            &s_dyt[(size_t)((ng * 2 + t) * 16 + ln) * WRW_DPITCH + r0]);
#pragma unroll
      for (int tap = 0; tap < 9; ++tap) {
        int dyy = tap / 3, dxx = tap % 3;
        bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
            &s_xt[(size_t)dxx * 64 * WRW_SPITCH + wrw_swz(mg * 16 + ln, (line + dyy) * LP + col)]);
#pragma unroll
        for (int t = 0; t < 2; ++t)
          acc[t][tap] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag[t], acc[t][tap], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  // partial[split][tap][c][k] bf16 (fp32 accumulated in regs, rounded once)
  bf16* base = partial + (int64_t)blockIdx.x * 9 * 64 * 64;
#pragma unroll
  for (int t = 0; t < 2; ++t)
#pragma unroll
    for (int tap = 0; tap < 9; ++tap)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int c = mg * 16 + km * 4 + r;
        int k = (ng * 2 + t) * 16 + ln;
        base[((int64_t)tap * 64 + c) * 64 + k] = (bf16)acc[t][tap][r];
      }
}

// Two-stage split combine. A single 144-block pass serially reading all 256
// strided splits left most of the chip idle (~25 us — dominated the call);
// stage 1 fans the split dim across gridDim.y, stage 2 folds the remainder
// and scatters the 64x64 slice into the full [K,C,3,3] channels_last dW.
#define WRW_SG 16  // split groups in stage 1
__global__ __launch_bounds__(256) void conv3x3_wrw_combine1_kernel(
    const bf16* __restrict__ partial, float* __restrict__ mid, int total, int splits) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total) return;
  int per = splits / WRW_SG;
  float s = 0.f;
  for (int sp = blockIdx.y * per; sp < (blockIdx.y + 1) * per; ++sp)
    s += (float)partial[(int64_t)sp * total + idx];
  mid[(int64_t)blockIdx.y * total + idx] = s;
}

__global__ __launch_bounds__(256) void conv3x3_wrw_combine2_kernel(
    const float* __restrict__ mid, bf16* __restrict__ dw_cl, int CT, int KT, int c0, int k0) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  const int total = 9 * 64 * 64;
  if (idx >= total) return;
  float s = 0.f;
#pragma unroll
  for (int g = 0; g < WRW_SG; ++g) s += mid[(int64_t)g * total + idx];
  int tap = idx / (64 * 64);
  int c = (idx / 64) % 64;
  int k = idx % 64;
  dw_cl[((int64_t)(k0 + k) * 9 + tap) * CT + c0 + c] = (bf16)s;
}

extern "C" void launch_conv3x3_wrw(const void* x, const void* dy, void* partial, float* mid,
                                   void* dw_cl, int Nn, int H, int W, int C, int K,
                                   hipStream_t s) {
  const int total = 9 * 64 * 64;
  dim3 g1((total + 255) / 256, WRW_SG, 1);
  dim3 g2((total + 255) / 256, 1, 1);
  for (int c0 = 0; c0 < C; c0 += 64)
    for (int k0 = 0; k0 < K; k0 += 64) {
      if (W == 32) {
        int n_tiles = Nn * (H / 4);
        conv3x3_wrw_kernel<32, 4><<<dim3(WRW_SPLITS, 1, 1), 512, 0, s>>>(
            (const bf16*)x, (const bf16*)dy, (bf16*)partial, Nn, H, C, c0, K, k0, n_tiles);
      } else {
        int n_tiles = Nn * (H / 8);
        conv3x3_wrw_kernel<16, 8><<<dim3(WRW_SPLITS, 1, 1), 512, 0, s>>>(
            (const bf16*)x, (const bf16*)dy, (bf16*)partial, Nn, H, C, c0, K, k0, n_tiles);
      }
      conv3x3_wrw_combine1_kernel<<<g1, 256, 0, s>>>((const bf16*)partial, mid, total,
                                                     WRW_SPLITS);
      conv3x3_wrw_combine2_kernel<<<g2, 256, 0, s>>>(mid, (bf16*)dw_cl, C, K, c0, k0);
    }
}
