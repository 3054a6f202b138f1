// Fused cosine-scores + streaming top-k: the Membrane salience recall
// kernel.
//
// Replaces the reference Membrane plugin's salience retrieval (external
// repo; config surface in brainplex configurator.ts:137-148) with an
// LDS-tiled MFMA scan over the HBM-resident embedding shard: for each
// 128-query block the kernel walks its swath of the [N, D] bf16 index,
// computes a 128x128 score tile per step (queries and index rows are
// L2-normalized, so dot = cosine), and maintains per-query top-k lists in
// LDS with a running-threshold filter — scores are NEVER materialized to
// HBM (a 4096 x 50M f32 score matrix would be 800 GB of traffic per
// step). A second small kernel merges the per-swath candidate lists.
//
// Grid: (Q/128) x n_swaths. Each block scans N/n_swaths index rows.
// LDS: staging tiles + score tile + top-k lists ~= 100 KB -> 1 block/CU;
// the 16 independent MFMA accumulators keep the matrix pipe busy at one
// wave/SIMD.
#include "common.hpp"

#define BM 128
#define BN 128
#define BK 32
#define PAD 8
#define LDS_STRIDE (BK + PAD)
#define TK_THREADS 256
#define TOPK_MAX 32

DEVINL void stage_tile_tk(const bf16* __restrict__ src, int ld, int row0,
                          int rows, int k0, bf16* lds) {
  int tid = threadIdx.x;
  int r = tid >> 1;
  int half = (tid & 1) * 16;
  bf16x8 v0 = {}, v1 = {};
  int gr = row0 + r;
  if (gr < rows) {
    const bf16* p = src + (size_t)gr * ld + k0 + half;
    v0 = *(const bf16x8*)(p);
    v1 = *(const bf16x8*)(p + 8);
  }
  *(bf16x8*)(lds + r * LDS_STRIDE + half) = v0;
  *(bf16x8*)(lds + r * LDS_STRIDE + half + 8) = v1;
}

// candidates layout: [n_qblocks][n_swaths][BM][k] for scores f32 and ids i32
extern "C" __global__ void __launch_bounds__(TK_THREADS)
topk_recall_kernel(const bf16* __restrict__ Q, const bf16* __restrict__ X,
                   int nq, int nx, int D, int k, int n_swaths,
                   float* __restrict__ cand_scores,
                   int32_t* __restrict__ cand_ids) {
  __shared__ bf16 Qs[2][BM * LDS_STRIDE];
  __shared__ bf16 Xs[2][BN * LDS_STRIDE];
  __shared__ float scores[BM][BN + 1];        // +1: avoid column-bank alignment
  __shared__ float topk_s[BM][TOPK_MAX];
  __shared__ int32_t topk_i[BM][TOPK_MAX];
  __shared__ float row_min[BM];
  __shared__ int row_min_slot[BM];

  int qb = blockIdx.x;
  int swath = blockIdx.y;
  int row0 = qb * BM;

  // swath range over the index
  long long per = ((long long)nx + n_swaths - 1) / n_swaths;
  long long x_begin = (long long)swath * per;
  long long x_end = min((long long)nx, x_begin + per);

  // init top-k lists
  for (int i = threadIdx.x; i < BM * k; i += blockDim.x) {
    topk_s[i / k][i % k] = -1e30f;
    topk_i[i / k][i % k] = -1;
  }
  for (int i = threadIdx.x; i < BM; i += blockDim.x) {
    row_min[i] = -1e30f;
    row_min_slot[i] = 0;
  }
  __syncthreads();

  int wid = wave_id();
  int wm = wid >> 1, wn = wid & 1;
  int lane = lane_id();
  int lrow = lane & 15;
  int kgrp = (lane >> 4) * 8;
  int nk = D / BK;

  for (long long x0 = x_begin; x0 < x_end; x0 += BN) {
    f32x4 acc[4][4] = {};
    stage_tile_tk(Q, D, row0, nq, 0, Qs[0]);
    stage_tile_tk(X, D, (int)x0, (int)x_end, 0, Xs[0]);
    __syncthreads();
    for (int kt = 0; kt < nk; ++kt) {
      int cur = kt & 1, nxt = cur ^ 1;
      if (kt + 1 < nk) {
        stage_tile_tk(Q, D, row0, nq, (kt + 1) * BK, Qs[nxt]);
        stage_tile_tk(X, D, (int)x0, (int)x_end, (kt + 1) * BK, Xs[nxt]);
      }
      bf16x8 qf[4], xf[4];
#pragma unroll
      for (int m = 0; m < 4; ++m)
        qf[m] = *(const bf16x8*)(Qs[cur] + (wm * 64 + m * 16 + lrow) * LDS_STRIDE + kgrp);
#pragma unroll
      for (int n = 0; n < 4; ++n)
        xf[n] = *(const bf16x8*)(Xs[cur] + (wn * 64 + n * 16 + lrow) * LDS_STRIDE + kgrp);
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[m], xf[n], acc[m][n], 0, 0, 0);
      __syncthreads();
    }

    // scores -> LDS (C/D map: col=lane&15, row=(lane>>4)*4+reg)
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        int col = wn * 64 + n * 16 + lrow;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = wm * 64 + m * 16 + (lane >> 4) * 4 + r;
          scores[row][col] = acc[m][n][r];
        }
      }
    __syncthreads();

    // streaming top-k update: each wave owns 32 rows; per row the wave's
    // 64 lanes scan 128 scores (2 each) against the row threshold, then
    // lane 0 serially inserts the (rare) survivors.
    for (int rr = 0; rr < 32; ++rr) {
      int row = wid * 32 + rr;
      float th = row_min[row];
      float s0 = scores[row][lane];
      float s1 = scores[row][lane + 64];
      bool valid0 = (x0 + lane) < x_end;
      bool valid1 = (x0 + lane + 64) < x_end;
      bool c0 = valid0 && s0 > th;
      bool c1 = valid1 && s1 > th;
      unsigned long long b0 = __ballot(c0);
      unsigned long long b1 = __ballot(c1);
      if (b0 == 0 && b1 == 0) continue;
      if (lane == 0) {
        // serial insert by lane 0 via LDS scan (rare path)
        for (int part = 0; part < 2; ++part) {
          unsigned long long bits = part == 0 ? b0 : b1;
          while (bits) {
            int src_lane = __ffsll((long long)bits) - 1;
            bits &= bits - 1;
            int col = src_lane + part * 64;
            float sv = scores[row][col];
            float mn = row_min[row];
            if (sv > mn) {
              int slot = row_min_slot[row];
              topk_s[row][slot] = sv;
              topk_i[row][slot] = (int32_t)(x0 + col);
              // recompute min
              float new_mn = topk_s[row][0];
              int new_slot = 0;
              for (int j = 1; j < k; ++j)
                if (topk_s[row][j] < new_mn) { new_mn = topk_s[row][j]; new_slot = j; }
              row_min[row] = new_mn;
              row_min_slot[row] = new_slot;
            }
          }
        }
      }
    }
    __syncthreads();
  }

  // write candidates: [qb][swath][row][k]
  size_t base = (((size_t)qb * n_swaths) + swath) * BM * k;
  for (int i = threadIdx.x; i < BM * k; i += blockDim.x) {
    cand_scores[base + i] = topk_s[i / k][i % k];
    cand_ids[base + i] = topk_i[i / k][i % k];
  }
}

// Merge per-swath candidates into final [nq][k] (one wave per query).
extern "C" __global__ void topk_merge_kernel(
    const float* __restrict__ cand_scores, const int32_t* __restrict__ cand_ids,
    int nq, int k, int n_swaths, float* __restrict__ out_scores,
    int32_t* __restrict__ out_ids) {
  int q = blockIdx.x * (blockDim.x / WAVE) + wave_id();
  if (q >= nq) return;
  int lane = lane_id();
  int qb = q / BM, row = q % BM;
  int total = n_swaths * k;  // candidates for this query
  // lane-local partial top-k via serial selection in registers:
  // simple approach: k rounds of argmax over remaining (total <= 32*16=512)
  // Each lane holds ceil(total/64) candidates.
  float my_s[16];
  int32_t my_i[16];
  int per_lane = (total + WAVE - 1) / WAVE;
  for (int j = 0; j < per_lane; ++j) {
    int idx = j * WAVE + lane;
    if (idx < total) {
      int sw = idx / k, slot = idx % k;
      size_t base = (((size_t)qb * n_swaths) + sw) * BM * k + (size_t)row * k;
      my_s[j] = cand_scores[base + slot];
      my_i[j] = cand_ids[base + slot];
    } else {
      my_s[j] = -1e30f;
      my_i[j] = -1;
    }
  }
  for (int sel = 0; sel < k; ++sel) {
    // local max
    float best = -1e30f;
    int bj = -1;
    for (int j = 0; j < per_lane; ++j)
      if (my_s[j] > best) { best = my_s[j]; bj = j; }
    // wave max reduce
    float wbest = best;
    int wlane = lane;
    for (int off = 32; off; off >>= 1) {
      float o = __shfl_down(wbest, off);
      int ol = __shfl_down(wlane, off);
      if (o > wbest) { wbest = o; wlane = ol; }
    }
    wbest = __shfl(wbest, 0);
    wlane = __shfl(wlane, 0);
    if (lane == wlane && bj >= 0) {
      if (lane == 0 || true) {
        // winner lane writes and retires its candidate
      }
      out_scores[(size_t)q * k + sel] = wbest;
      out_ids[(size_t)q * k + sel] = my_i[bj];
      my_s[bj] = -1e30f;
    }
    __syncthreads();
  }
}
