// Batched Levenshtein edit distance — the cortex doom-loop detector's
// exec-command similarity (reference doom-loop.ts:76-90
// computes Levenshtein on <=500-char commands per chain, host-side; here
// it is a wavefront-DP kernel so a whole batch of candidate pairs runs
// in one launch).
//
// One 64-lane wave per string pair, anti-diagonal dynamic programming:
// cells on a diagonal are independent, so the wave sweeps each diagonal
// in ceil(cells/64) rounds, keeping the last two diagonals in LDS.
// Strings are capped at ED_MAX bytes (the reference caps at 500).
#include "common.hpp"

#define ED_MAX 512
#define ED_THREADS 64

extern "C" __global__ void __launch_bounds__(ED_THREADS)
edit_distance_kernel(const uint8_t* __restrict__ bytes,
                     const int32_t* __restrict__ offsets,  // [2*n_pairs + 1]
                     int32_t* __restrict__ out, int n_pairs) {
  __shared__ uint16_t diag[3][ED_MAX + 1];
  __shared__ uint8_t sa[ED_MAX], sb[ED_MAX];

  int pair = blockIdx.x;
  if (pair >= n_pairs) return;
  int lane = threadIdx.x;

  long long a0 = offsets[2 * pair], a1 = offsets[2 * pair + 1];
  long long b0 = offsets[2 * pair + 1], b1 = offsets[2 * pair + 2];
  int la = (int)min((long long)ED_MAX, a1 - a0);
  int lb = (int)min((long long)ED_MAX, b1 - b0);

  for (int i = lane; i < la; i += ED_THREADS) sa[i] = bytes[a0 + i];
  for (int j = lane; j < lb; j += ED_THREADS) sb[j] = bytes[b0 + j];
  __syncthreads();

  if (la == 0 || lb == 0) {
    if (lane == 0) out[pair] = la + lb;
    return;
  }

  // diag d holds D[i][d-i]; rotate three buffers
  int cur = 0;
  for (int d = 0; d <= la + lb; ++d) {
    int prev = (cur + 2) % 3, prev2 = (cur + 1) % 3;
    int ilo = max(0, d - lb), ihi = min(la, d);
    for (int i = ilo + lane; i <= ihi; i += ED_THREADS) {
      int j = d - i;
      uint16_t v;
      if (i == 0) {
        v = (uint16_t)j;
      } else if (j == 0) {
        v = (uint16_t)i;
      } else {
        uint32_t del = diag[prev][i - 1];      // D[i-1][j]
        uint32_t ins = diag[prev][i];          // D[i][j-1]
        uint32_t sub = diag[prev2][i - 1];     // D[i-1][j-1]
        uint32_t v1 = sub + ((sa[i - 1] != sb[j - 1]) ? 1u : 0u);
        uint32_t v2 = (del < ins ? del : ins) + 1u;
        v = (uint16_t)(v1 < v2 ? v1 : v2);
      }
      diag[cur][i] = v;
    }
    __syncthreads();
    cur = (cur + 1) % 3;
  }
  if (lane == 0) out[pair] = diag[(cur + 2) % 3][la];
}
