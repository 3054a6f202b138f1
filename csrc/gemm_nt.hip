// bf16 NT GEMM with fused epilogue: C[M,N] = A[M,K] @ B[N,K]^T (+bias, act).
//
// The MI355X-native classifier/tagger head (replacing the reference's
// remote-LLM / regex-only scoring paths): features from csrc/encoder.hip
// times a random-init head [C_out, K]. Both operands are K-contiguous
// row-major, which is the natural MFMA feed (no transposes).
//
// Structure: the canonical CDNA4 tile (cdna_hip_programming.md §5):
// 128x128 block tile, BK=32, 4 waves in a 2x2 grid each owning a 64x64
// sub-tile as 4x4 fragments of v_mfma_f32_16x16x32_bf16, LDS double
// buffered with +8 bf16 row padding against ds_read_b128 bank conflicts.
// Correctness-first version; the deep-pipelined 256^2 8-phase schedule is
// a later optimization pass once rocprof baselines exist.
#include "common.hpp"

#define BM 128
#define BN 128
#define BK 32
#define PAD 8  // bf16 elements of row padding (16 B): distinct banks per row
#define LDS_STRIDE (BK + PAD)
#define GEMM_THREADS 256

// Epilogue activation codes
#define ACT_NONE 0
#define ACT_SIGMOID 1
#define ACT_RELU 2

DEVINL float apply_act(float v, int act) {
  if (act == ACT_SIGMOID) return 1.0f / (1.0f + __expf(-v));
  if (act == ACT_RELU) return fmaxf(v, 0.0f);
  return v;
}

// Cooperative load of a [128 rows x 32 cols] bf16 tile into padded LDS.
// Each of the 256 threads moves 16 contiguous bf16 (32 B): row = tid/2,
// half = tid%2. Rows past `rows` are zero-filled.
DEVINL void stage_tile(const bf16* __restrict__ src, int ld, int row0,
                       int rows, int k0, int K, bf16* lds) {
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

extern "C" __global__ void __launch_bounds__(GEMM_THREADS)
gemm_nt_bf16_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                    float* __restrict__ C_f32, bf16* __restrict__ C_bf16,
                    const float* __restrict__ bias, int M, int N, int K,
                    int act, int out_bf16) {
  __shared__ bf16 As[2][BM * LDS_STRIDE];
  __shared__ bf16 Bs[2][BN * LDS_STRIDE];

  int bm = blockIdx.x;  // M tile
  int bn = blockIdx.y;  // N tile
  int row0 = bm * BM, col0 = bn * BN;

  int wid = wave_id();          // 0..3
  int wm = wid >> 1, wn = wid & 1;  // 2x2 wave grid, each 64x64
  int lane = lane_id();
  int lrow = lane & 15;         // fragment row/col lane index
  int kgrp = (lane >> 4) * 8;   // fragment k offset

  f32x4 acc[4][4] = {};

  stage_tile(A, K, row0, M, 0, K, As[0]);
  stage_tile(B, K, col0, N, 0, K, Bs[0]);
  __syncthreads();

  int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1, nxt = cur ^ 1;
    if (kt + 1 < nk) {
      stage_tile(A, K, row0, M, (kt + 1) * BK, K, As[nxt]);
      stage_tile(B, K, col0, N, (kt + 1) * BK, K, Bs[nxt]);
    }
    // two K sub-steps of 32 -> wait, BK==32 is one MFMA K per fragment set
    bf16x8 afrag[4], bfrag[4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
      afrag[m] = *(const bf16x8*)(As[cur] + (wm * 64 + m * 16 + lrow) * LDS_STRIDE + kgrp);
#pragma unroll
    for (int n = 0; n < 4; ++n)
      bfrag[n] = *(const bf16x8*)(Bs[cur] + (wn * 64 + n * 16 + lrow) * LDS_STRIDE + kgrp);
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
    __syncthreads();
  }

  // epilogue: C/D layout col=lane&15, row=(lane>>4)*4+reg (guide §3)
  int crow_base = row0 + wm * 64;
  int ccol_base = col0 + wn * 64;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int col = ccol_base + n * 16 + lrow;
      if (col >= N) continue;
      float bv = bias ? bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow_base + m * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        float v = apply_act(acc[m][n][r] + bv, act);
        if (out_bf16)
          C_bf16[(size_t)row * N + col] = bf16(v);
        else
          C_f32[(size_t)row * N + col] = v;
      }
    }
  }
}
