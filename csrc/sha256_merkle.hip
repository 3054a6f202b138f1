// Batched SHA-256 + Merkle tree for the audit trail hot path.
//
// Replaces the reference's per-record JS hashing (governance
// audit-trail.ts + the Merkle chain the governance README promises) with
// one kernel over the whole message batch: leaf digests of
// variable-length records, then pairwise tree levels (odd node
// duplicated), matching vainplex_openclaw_amd.governance.audit.merkle_root
// bit for bit. Per-batch roots are all-reduced across GPUs over RCCL.
//
// One thread per record; a 4096-record batch is 64 waves — the kernel is
// latency-bound on the record bytes (L2-resident), ~microseconds.
#include "common.hpp"

__constant__ uint32_t K256[64] = {
    0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b, 0x59f111f1,
    0x923f82a4, 0xab1c5ed5, 0xd807aa98, 0x12835b01, 0x243185be, 0x550c7dc3,
    0x72be5d74, 0x80deb1fe, 0x9bdc06a7, 0xc19bf174, 0xe49b69c1, 0xefbe4786,
    0x0fc19dc6, 0x240ca1cc, 0x2de92c6f, 0x4a7484aa, 0x5cb0a9dc, 0x76f988da,
    0x983e5152, 0xa831c66d, 0xb00327c8, 0xbf597fc7, 0xc6e00bf3, 0xd5a79147,
    0x06ca6351, 0x14292967, 0x27b70a85, 0x2e1b2138, 0x4d2c6dfc, 0x53380d13,
    0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85, 0xa2bfe8a1, 0xa81a664b,
    0xc24b8b70, 0xc76c51a3, 0xd192e819, 0xd6990624, 0xf40e3585, 0x106aa070,
    0x19a4c116, 0x1e376c08, 0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a,
    0x5b9cca4f, 0x682e6ff3, 0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208,
    0x90befffa, 0xa4506ceb, 0xbef9a3f7, 0xc67178f2};

DEVINL uint32_t rotr(uint32_t x, int n) { return (x >> n) | (x << (32 - n)); }

struct Sha256Ctx {
  uint32_t h[8];
  DEVINL void init() {
    h[0] = 0x6a09e667; h[1] = 0xbb67ae85; h[2] = 0x3c6ef372; h[3] = 0xa54ff53a;
    h[4] = 0x510e527f; h[5] = 0x9b05688c; h[6] = 0x1f83d9ab; h[7] = 0x5be0cd19;
  }
  DEVINL void block(const uint8_t* p) {
    uint32_t w[16];
#pragma unroll
    for (int i = 0; i < 16; ++i)
      w[i] = (uint32_t(p[4 * i]) << 24) | (uint32_t(p[4 * i + 1]) << 16) |
             (uint32_t(p[4 * i + 2]) << 8) | uint32_t(p[4 * i + 3]);
    uint32_t a = h[0], b = h[1], c = h[2], d = h[3];
    uint32_t e = h[4], f = h[5], g = h[6], hh = h[7];
#pragma unroll
    for (int i = 0; i < 64; ++i) {
      uint32_t wi;
      if (i < 16) {
        wi = w[i];
      } else {
        uint32_t s0 = rotr(w[(i + 1) & 15], 7) ^ rotr(w[(i + 1) & 15], 18) ^ (w[(i + 1) & 15] >> 3);
        uint32_t s1 = rotr(w[(i + 14) & 15], 17) ^ rotr(w[(i + 14) & 15], 19) ^ (w[(i + 14) & 15] >> 10);
        wi = w[i & 15] + s0 + w[(i + 9) & 15] + s1;
        w[i & 15] = wi;
      }
      uint32_t S1 = rotr(e, 6) ^ rotr(e, 11) ^ rotr(e, 25);
      uint32_t ch = (e & f) ^ (~e & g);
      uint32_t t1 = hh + S1 + ch + K256[i] + wi;
      uint32_t S0 = rotr(a, 2) ^ rotr(a, 13) ^ rotr(a, 22);
      uint32_t mj = (a & b) ^ (a & c) ^ (b & c);
      uint32_t t2 = S0 + mj;
      hh = g; g = f; f = e; e = d + t1;
      d = c; c = b; b = a; a = t1 + t2;
    }
    h[0] += a; h[1] += b; h[2] += c; h[3] += d;
    h[4] += e; h[5] += f; h[6] += g; h[7] += hh;
  }
  // full message hash (handles padding); len in bytes
  DEVINL void hash(const uint8_t* data, uint32_t len) {
    init();
    uint32_t full = len / 64;
    for (uint32_t i = 0; i < full; ++i) block(data + 64 * i);
    uint8_t tail[128];
    uint32_t rem = len - full * 64;
#pragma unroll 4
    for (uint32_t i = 0; i < rem; ++i) tail[i] = data[full * 64 + i];
    tail[rem] = 0x80;
    uint32_t padded = (rem + 9 <= 64) ? 64 : 128;
    for (uint32_t i = rem + 1; i < padded - 8; ++i) tail[i] = 0;
    uint64_t bits = uint64_t(len) * 8;
#pragma unroll
    for (int i = 0; i < 8; ++i) tail[padded - 1 - i] = uint8_t(bits >> (8 * i));
    block(tail);
    if (padded == 128) block(tail + 64);
  }
  DEVINL void digest(uint8_t* out) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      out[4 * i] = uint8_t(h[i] >> 24);
      out[4 * i + 1] = uint8_t(h[i] >> 16);
      out[4 * i + 2] = uint8_t(h[i] >> 8);
      out[4 * i + 3] = uint8_t(h[i]);
    }
  }
};

// Leaf digests: one thread per variable-length record.
extern "C" __global__ void sha256_leaves_kernel(
    const uint8_t* __restrict__ bytes, const int32_t* __restrict__ offsets,
    uint8_t* __restrict__ digests, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int32_t lo = offsets[i], hi = offsets[i + 1];
  Sha256Ctx ctx;
  ctx.hash(bytes + lo, uint32_t(hi - lo));
  ctx.digest(digests + 32 * i);
}

// One Merkle level: out[i] = sha256(in[2i] || in[2i+1]); odd node duplicated.
extern "C" __global__ void merkle_level_kernel(
    const uint8_t* __restrict__ in, uint8_t* __restrict__ out, int n_in) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  int n_out = (n_in + 1) / 2;
  if (i >= n_out) return;
  uint8_t buf[64];
  const uint8_t* left = in + 64 * i;  // pairs are contiguous: 2i, 2i+1
#pragma unroll
  for (int k = 0; k < 32; ++k) buf[k] = in[32 * (2 * i) + k];
  int right = (2 * i + 1 < n_in) ? 2 * i + 1 : 2 * i;
#pragma unroll
  for (int k = 0; k < 32; ++k) buf[32 + k] = in[32 * right + k];
  (void)left;
  Sha256Ctx ctx;
  ctx.hash(buf, 64);
  ctx.digest(out + 32 * i);
}
