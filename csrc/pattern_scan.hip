// Multi-pattern DFA scan over a message batch.
//
// Replaces the reference's per-message regex loops (redaction
// registry.ts findMatches, claim-detector.ts, knowledge-engine
// patterns.ts, governance policy regex conditions) with one kernel: each
// thread walks the packed DFA tables (vainplex_openclaw_amd/ops/dfa.py
// MultiDFA.pack) over its message and ORs per-state accept masks into a
// u64 hit mask per message.
//
// The tables (few hundred KB) stay L2-resident; the per-family byte->class
// maps are staged in LDS. The walk is a dependent-load chain, so we give
// every thread its own message and rely on wave-level parallelism (a
// 4096-message batch = 64 waves spread over the 256 CUs).
#include "common.hpp"

// meta row: (next_base in u16 units, state_base, n_classes, class_map_row)
extern "C" __global__ void dfa_scan_kernel(
    const uint8_t* __restrict__ bytes, const int32_t* __restrict__ offsets,
    const uint16_t* __restrict__ next_tab, const unsigned long long* __restrict__ accept,
    const unsigned long long* __restrict__ eof_mask,
    const uint8_t* __restrict__ class_maps, const int32_t* __restrict__ meta,
    int n_dfas, unsigned long long* __restrict__ hits, int n_msgs) {
  extern __shared__ uint8_t lds_class[];  // n_dfas * 256
  for (int i = threadIdx.x; i < n_dfas * 256; i += blockDim.x)
    lds_class[i] = class_maps[i];
  __syncthreads();

  int mi = blockIdx.x * blockDim.x + threadIdx.x;
  if (mi >= n_msgs) return;
  int32_t lo = offsets[mi], hi = offsets[mi + 1];
  unsigned long long mask = 0ull;
  for (int d = 0; d < n_dfas; ++d) {
    int32_t next_base = meta[4 * d + 0];
    int32_t state_base = meta[4 * d + 1];
    int32_t ncls = meta[4 * d + 2];
    const uint8_t* cmap = lds_class + 256 * meta[4 * d + 3];
    uint32_t state = 0;
    mask |= accept[state_base];
    for (int32_t p = lo; p < hi; ++p) {
      uint32_t cls = cmap[bytes[p]];
      state = next_tab[next_base + state * ncls + cls];
      mask |= accept[state_base + state];
    }
    mask |= eof_mask[state_base + state];
  }
  hits[mi] = mask;
}

// Variant scanning ALL families in one launch: families are just separate
// packed MultiDFAs laid out back to back; each writes its own hit column.
// hits layout: [n_families, n_msgs].
extern "C" __global__ void dfa_scan_multi_kernel(
    const uint8_t* __restrict__ bytes, const int32_t* __restrict__ offsets,
    const uint16_t* __restrict__ next_tab, const unsigned long long* __restrict__ accept,
    const unsigned long long* __restrict__ eof_mask,
    const uint8_t* __restrict__ class_maps, const int32_t* __restrict__ meta,
    const int32_t* __restrict__ family_dfa_ranges,  // [n_families, 2] (begin, end) into meta rows
    int n_families, unsigned long long* __restrict__ hits, int n_msgs,
    int n_dfas_total) {
  extern __shared__ uint8_t lds_class[];
  for (int i = threadIdx.x; i < n_dfas_total * 256; i += blockDim.x)
    lds_class[i] = class_maps[i];
  __syncthreads();

  int mi = blockIdx.x * blockDim.x + threadIdx.x;
  if (mi >= n_msgs) return;
  int32_t lo = offsets[mi], hi = offsets[mi + 1];
  for (int f = 0; f < n_families; ++f) {
    int32_t dbeg = family_dfa_ranges[2 * f], dend = family_dfa_ranges[2 * f + 1];
    unsigned long long mask = 0ull;
    for (int d = dbeg; d < dend; ++d) {
      int32_t next_base = meta[4 * d + 0];
      int32_t state_base = meta[4 * d + 1];
      int32_t ncls = meta[4 * d + 2];
      const uint8_t* cmap = lds_class + 256 * meta[4 * d + 3];
      uint32_t state = 0;
      mask |= accept[state_base];
      for (int32_t p = lo; p < hi; ++p) {
        uint32_t cls = cmap[bytes[p]];
        state = next_tab[next_base + state * ncls + cls];
        mask |= accept[state_base + state];
      }
      mask |= eof_mask[state_base + state];
    }
    hits[(size_t)f * n_msgs + mi] = mask;
  }
}
