// GPU hash-table probe of the fact registry (claims stage).
//
// Reference semantics: fact-checker.ts:67-123 keeps a
// Map<"subject|predicate" -> Fact> and checks detected claims against it
// (verdicts verified / contradicted / unverified). The batched MI355X
// form: the host packs the registry into an open-addressing table of
// (key, value) u64 pairs where
//   key   = fnv1a64(lower(subject) + "|" + predicate)
//   value = fnv1a64(lower(object))
// and this kernel, for every message whose claims-DFA mask hit, probes
// the table with key = mix(token_hash, predicate_hash(claim family))
// for each word token, comparing the stored object hash against the
// other tokens of the message:
//   table hit + object token present  -> verified
//   table hit + object token absent   -> contradicted
//   no table hit                      -> unverified (not counted)
// This is the reference's subject|predicate lookup restricted to
// single-token subjects/objects (the synthetic-fact and demo shapes);
// multi-word subjects fall back to the host FactRegistry path.
//
// One thread per message (dependent-load walk like the DFA scan); the
// table lives in L2/HBM, token hashes are staged in a per-thread local
// array (<= 96 tokens of a 500 B message).
#include "common.hpp"

#define FNV_OFFSET 0xCBF29CE484222325ull
#define FNV_PRIME 1099511628211ull
#define MAX_TOKENS 96

DEVINL bool is_word_byte(uint8_t b) {
  return (b >= 'a' && b <= 'z') || (b >= 'A' && b <= 'Z') ||
         (b >= '0' && b <= '9') || b == '_' || b == '-' || b == '.';
}

DEVINL uint8_t lower_byte(uint8_t b) {
  return (b >= 'A' && b <= 'Z') ? (uint8_t)(b + 32) : b;
}

DEVINL unsigned long long mix_key(unsigned long long subject_h,
                                  unsigned long long predicate_h) {
  // continue the subject FNV stream through '|' then fold in the
  // predicate hash bytes — matches the host build_fact_table mix
  unsigned long long h = subject_h;
  h = (h ^ (unsigned long long)'|') * FNV_PRIME;
  for (int i = 0; i < 8; ++i) {
    h = (h ^ ((predicate_h >> (8 * i)) & 0xffull)) * FNV_PRIME;
  }
  return h;
}

extern "C" __global__ void fact_probe_kernel(
    const uint8_t* __restrict__ bytes, const int32_t* __restrict__ offsets,
    const unsigned long long* __restrict__ claims_mask,   // [B]
    const unsigned long long* __restrict__ pred_hash,     // [64] per claim bit (0 = unused)
    const unsigned long long* __restrict__ table_keys,    // [S] 0 = empty
    const unsigned long long* __restrict__ table_vals,    // [S]
    int table_pow2,                                        // S = 1 << table_pow2
    int32_t* __restrict__ verified, int32_t* __restrict__ contradicted,
    int n_msgs) {
  int mi = blockIdx.x * blockDim.x + threadIdx.x;
  if (mi >= n_msgs) return;
  unsigned long long cmask = claims_mask[mi];
  if (cmask == 0ull) { verified[mi] = 0; contradicted[mi] = 0; return; }

  // tokenize once: FNV-1a of each lowercased word token (len >= 2)
  unsigned long long tok[MAX_TOKENS];
  int n_tok = 0;
  int32_t lo = offsets[mi], hi = offsets[mi + 1];
  unsigned long long h = FNV_OFFSET;
  int tlen = 0;
  for (int32_t p = lo; p <= hi; ++p) {
    uint8_t b = (p < hi) ? bytes[p] : (uint8_t)' ';
    if (is_word_byte(b)) {
      h = (h ^ (unsigned long long)lower_byte(b)) * FNV_PRIME;
      ++tlen;
    } else if (tlen > 0) {
      if (tlen >= 2 && n_tok < MAX_TOKENS) tok[n_tok++] = h;
      h = FNV_OFFSET;
      tlen = 0;
    }
  }

  const unsigned long long mask_S = (1ull << table_pow2) - 1ull;
  int n_ver = 0, n_con = 0;
  for (int bit = 0; bit < 64; ++bit) {
    if (!((cmask >> bit) & 1ull)) continue;
    unsigned long long ph = pred_hash[bit];
    if (ph == 0ull) continue;
    for (int t = 0; t < n_tok; ++t) {
      unsigned long long key = mix_key(tok[t], ph);
      if (key == 0ull) key = 1ull;  // 0 is the empty slot marker
      unsigned long long slot = key & mask_S;
      unsigned long long val = 0ull;
      bool found = false;
      for (int probe = 0; probe < 16; ++probe) {
        unsigned long long k = table_keys[slot];
        if (k == 0ull) break;
        if (k == key) { val = table_vals[slot]; found = true; break; }
        slot = (slot + 1ull) & mask_S;
      }
      if (!found) continue;
      bool obj_present = false;
      for (int u = 0; u < n_tok; ++u) {
        if (tok[u] == val) { obj_present = true; break; }
      }
      if (obj_present) ++n_ver; else ++n_con;
    }
  }
  verified[mi] = n_ver;
  contradicted[mi] = n_con;
}
