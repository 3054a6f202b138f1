// Fused cosine-scores + streaming top-k: the Membrane salience recall
// kernel (v2).
//
// Replaces the reference Membrane plugin's salience retrieval (external
// repo; config surface in brainplex configurator.ts:137-148) with an
// LDS-tiled MFMA scan over the HBM-resident embedding shard. Scores are
// never materialized to HBM (4096 x 50M f32 would be 800 GB of traffic
// per step): each 128-query block walks its swath of the [N, D] bf16
// index, computes 128x256 score tiles with v_mfma_f32_16x16x32_bf16 and
// filters them against per-row running top-k thresholds straight from the
// accumulator registers.
//
// Structure (cdna_hip_programming.md §5 "step 3"):
// - 8 waves (512 thr), block tile BM=128 x BN=256, BK=32, wave grid 2x4
//   (each wave a 64x64 subtile = 4x4 fragments, 16 independent
//   accumulators keep the matrix pipe busy at 2 waves/SIMD).
// - global_load_lds width 16 staging, LDS image lane-linear with the
//   XOR swizzle applied on the SOURCE k-group and again on the ds_read
//   offset (guide rule 21): slot ^= (row>>2)&3 makes the 16-lane
//   ds_read_b128 groups conflict-free without row padding.
// - double-buffered LDS, one barrier pair per K-step (the deep-pipelined
//   8-phase schedule is the next rung once this baseline is profiled).
// - top-k: per-row threshold in LDS; lanes test their 16 accumulator
//   values (C/D map row=(lane>>4)*4+r, col=lane&15), survivors go through
//   a bounded LDS candidate queue drained by row-owning lanes.
#include "common.hpp"

#define BM 128
#define BN 256
#define BK 32
#define TK_THREADS 512
#define TOPK_MAX 32
#define QCAP 2048

// LDS tile addressing: row-major [rows][BK] bf16, 64 B per row = 4 slots
// of 16 B. Swizzle: slot' = slot ^ ((row>>2)&3).
DEVINL uint32_t lds_off_bytes(uint32_t row, uint32_t slot) {
  return (row * 4u + (slot ^ ((row >> 2u) & 3u))) * 16u;
}

// Stage a [rows x BK] tile into LDS. Piece p (16 B) = row p/4, slot p%4;
// the SOURCE k-group is the swizzled slot (involution with the read side).
//
// USE_GLDS=1: global_load_lds — the LDS operand must be the WAVE-UNIFORM
// base of the wave's 64 consecutive pieces (the hardware adds lane*16;
// a per-lane destination silently scatters / faults). The per-lane part
// lives only in the global SOURCE address.
#ifndef USE_GLDS
#define USE_GLDS 1
#endif

DEVINL void stage_glds(const bf16* __restrict__ src, long long ld,
                       long long row0, long long row_max, int k0,
                       bf16* lds_base, int tile_rows) {
  int n_pieces = tile_rows * 4;  // 16B pieces in the tile
  int w = wave_id();
  int lane = lane_id();
  for (int piece0 = w * WAVE; piece0 < n_pieces; piece0 += (TK_THREADS / WAVE) * WAVE) {
    int piece = piece0 + lane;
    uint32_t r = piece >> 2;
    uint32_t slot = piece & 3;
    uint32_t src_slot = slot ^ ((r >> 2u) & 3u);
    long long gr = row0 + r;
    if (gr >= row_max) gr = row_max - 1;  // clamp: garbage filtered later
    const bf16* p = src + gr * ld + k0 + src_slot * 8;
#if USE_GLDS
    int piece0_u = __builtin_amdgcn_readfirstlane(piece0);
    auto ldst = (__attribute__((address_space(3))) char*)lds_base + piece0_u * 16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)p,
        (__attribute__((address_space(3))) void*)ldst, 16, 0, 0);
#else
    *(bf16x8*)((char*)lds_base + piece * 16) = *(const bf16x8*)p;
#endif
  }
}

extern "C" __global__ void __launch_bounds__(TK_THREADS)
topk_recall_kernel(const bf16* __restrict__ Q, const bf16* __restrict__ X,
                   int nq, int nx, int D, int k, int n_swaths,
                   float* __restrict__ cand_scores,
                   int32_t* __restrict__ cand_ids) {
  __shared__ bf16 lds_all[2 * (BM + BN) * BK];
#define QS(buf) (lds_all + (buf) * BM * BK)
#define XS(buf) (lds_all + 2 * BM * BK + (buf) * BN * BK)
  __shared__ float topk_s[BM][TOPK_MAX];
  __shared__ int32_t topk_i[BM][TOPK_MAX];
  __shared__ float row_min[BM];
  __shared__ int row_min_slot[BM];
  __shared__ float q_score[QCAP];
  __shared__ uint32_t q_meta[QCAP];  // (row<<16) | col_in_tile
  __shared__ int q_count;
  __shared__ int q_overflow;

  int qb = blockIdx.x;
  int swath = blockIdx.y;
  long long row0 = (long long)qb * BM;

  long long per = ((long long)nx + n_swaths - 1) / n_swaths;
  per = ((per + BN - 1) / BN) * BN;
  long long x_begin = (long long)swath * per;
  long long x_end = min((long long)nx, x_begin + per);

  for (int i = threadIdx.x; i < BM * TOPK_MAX; i += blockDim.x) {
    topk_s[i / TOPK_MAX][i % TOPK_MAX] = -1e30f;
    topk_i[i / TOPK_MAX][i % TOPK_MAX] = -1;
  }
  for (int i = threadIdx.x; i < BM; i += blockDim.x) {
    row_min[i] = -1e30f;
    row_min_slot[i] = 0;
  }
  if (threadIdx.x == 0) { q_count = 0; q_overflow = 0; }
  __syncthreads();

  int wid = wave_id();          // 0..7
  int wm = wid >> 2, wn = wid & 3;  // 2 x 4 wave grid
  int lane = lane_id();
  int lrow = lane & 15;
  int kslot = lane >> 4;        // 0..3 -> k-group (8 bf16 = 16 B)
  int nk = D / BK;

  for (long long x0 = x_begin; x0 < x_end; x0 += BN) {
    f32x4 acc[4][4] = {};
    stage_glds(Q, D, row0, nq, 0, QS(0), BM);
    stage_glds(X, D, x0, (long long)nx, 0, XS(0), BN);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    for (int kt = 0; kt < nk; ++kt) {
      int cur = kt & 1, nxt = cur ^ 1;
      if (kt + 1 < nk) {
        stage_glds(Q, D, row0, nq, (kt + 1) * BK, QS(nxt), BM);
        stage_glds(X, D, x0, (long long)nx, (kt + 1) * BK, XS(nxt), BN);
      }
      bf16x8 qf[4], xf[4];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        uint32_t r = wm * 64 + m * 16 + lrow;
        qf[m] = *(const bf16x8*)((const char*)QS(cur) + lds_off_bytes(r, kslot));
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        uint32_t r = wn * 64 + n * 16 + lrow;
        xf[n] = *(const bf16x8*)((const char*)XS(cur) + lds_off_bytes(r, kslot));
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[m], xf[n], acc[m][n], 0, 0, 0);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    }

    // ---- streaming top-k from the accumulators -------------------------
    // lane holds acc[m][n][r] at row = wm*64+m*16+(lane>>4)*4+r,
    //                         col = wn*64+n*16+(lane&15)
    // Rounds: push survivors into the queue; drain by row-owner lanes;
    // repeat if the queue overflowed (only plausible on the first tile).
    // pending bit (m*16 + n*4 + r): value not yet pushed/retired
    // STATIC m/n/r indexing only: a dynamic index into acc forces the
    // compiler to spill all 64 accumulator VGPRs to scratch, and every
    // MFMA then round-trips scratch memory (measured 46x slowdown).
    unsigned long long pending = ~0ull;
    for (int round = 0; ; ++round) {
      unsigned long long still = 0ull;
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int vi = m * 16 + n * 4 + r;
            if (!((pending >> vi) & 1ull)) continue;
            int row = wm * 64 + m * 16 + (lane >> 4) * 4 + r;
            if ((row0 + row) >= nq) continue;
            long long col = x0 + wn * 64 + n * 16 + (lane & 15);
            if (col >= x_end) continue;
            float v = acc[m][n][r];
            if (!(v > row_min[row])) continue;
            int idx = atomicAdd(&q_count, 1);
            if (idx < QCAP) {
              q_score[idx] = v;
              q_meta[idx] = (uint32_t(row) << 16) | uint32_t(col - x0);
            } else {
              still |= 1ull << vi;  // retry just this value next round
              atomicExch(&q_overflow, 1);
            }
          }
      __syncthreads();
      // drain: lanes 0..15 of each wave own rows wid*16 + (lane)
      int total = min(q_count, QCAP);
      if (lane < 16) {
        int my_row = wid * 16 + lane;
        for (int i = 0; i < total; ++i) {
          uint32_t meta = q_meta[i];
          int row = int(meta >> 16);
          if (row != my_row) continue;
          float v = q_score[i];
          if (v > row_min[row]) {
            int slot = row_min_slot[row];
            topk_s[row][slot] = v;
            topk_i[row][slot] = int32_t(x0 + (meta & 0xFFFFu));
            float mn = topk_s[row][0];
            int ms = 0;
            for (int j = 1; j < k; ++j)
              if (topk_s[row][j] < mn) { mn = topk_s[row][j]; ms = j; }
            row_min[row] = mn;
            row_min_slot[row] = ms;
          }
        }
      }
      __syncthreads();
      int of = q_overflow;
      __syncthreads();  // all reads of q_overflow done before the reset
      if (threadIdx.x == 0) { q_count = 0; q_overflow = 0; }
      pending = still;
      if (!of) break;   // of is uniform (LDS): no divergence
      __syncthreads();  // reset visible before next round's pushes
    }
    __syncthreads();
  }

  // candidates out: [qb][swath][row][k]
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
  int total = n_swaths * k;
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
    float best = -1e30f;
    int bj = -1;
    for (int j = 0; j < per_lane; ++j)
      if (my_s[j] > best) { best = my_s[j]; bj = j; }
    float wbest = best;
    int wlane = lane;
    for (int off = 32; off; off >>= 1) {
      float o = __shfl_down(wbest, off);
      int ol = __shfl_down(wlane, off);
      if (o > wbest) { wbest = o; wlane = ol; }
    }
    wbest = __shfl(wbest, 0);
    wlane = __shfl(wlane, 0);
    if (lane == wlane) {
      out_scores[(size_t)q * k + sel] = wbest;
      out_ids[(size_t)q * k + sel] = (bj >= 0) ? my_i[bj] : -1;
      if (bj >= 0) my_s[bj] = -1e30f;
    }
  }
}
