// Message encoder: byte 4-gram hash -> embedding gather -> mean-pool ->
// L2 normalize -> bf16 features.
//
// This is the framework's "tokenizer + embedding" for the batched
// firewall/membrane path: the classifier head (GEMM, csrc/gemm_nt.hip)
// and the Membrane recall queries (csrc/topk_recall.hip) both consume
// these features. Random-init embedding table (the bench has no network
// for checkpoints); the table layout is [vocab, dim] bf16, vocab a power
// of two so the hash is a mask.
//
// One block per message, 256 threads; threads cooperatively compute token
// hashes into LDS, then stride the embedding dim: thread t accumulates
// dims [4t, 4t+4) over all tokens with bf16x4 loads (coalesced across the
// block), so the gather streams at HBM/L2 line granularity.
#include "common.hpp"

#define ENC_THREADS 256
#define MAX_TOKENS 512

DEVINL uint32_t fnv1a4(uint8_t a, uint8_t b, uint8_t c, uint8_t d) {
  uint32_t h = 2166136261u;
  h = (h ^ a) * 16777619u;
  h = (h ^ b) * 16777619u;
  h = (h ^ c) * 16777619u;
  h = (h ^ d) * 16777619u;
  return h;
}

extern "C" __global__ void __launch_bounds__(ENC_THREADS)
encode_messages_kernel(const uint8_t* __restrict__ bytes,
                       const int32_t* __restrict__ offsets,
                       const bf16* __restrict__ embed, int vocab_mask,
                       int dim, bf16* __restrict__ out, int n_msgs,
                       int normalize) {
  __shared__ int32_t tok[MAX_TOKENS];
  __shared__ int32_t n_tok_sh;
  __shared__ float norm_sh;

  int mi = blockIdx.x;
  if (mi >= n_msgs) return;
  int32_t lo = offsets[mi], hi = offsets[mi + 1];
  int32_t len = hi - lo;
  // token stride: subsample long messages so tokens fit MAX_TOKENS
  int32_t n_pos = max(len - 3, 0);
  int32_t stride = (n_pos + MAX_TOKENS - 1) / MAX_TOKENS;
  if (stride < 1) stride = 1;
  int32_t n_tok = (n_pos + stride - 1) / stride;
  if (threadIdx.x == 0) n_tok_sh = n_tok;
  for (int t = threadIdx.x; t < n_tok; t += blockDim.x) {
    int32_t p = lo + t * stride;
    tok[t] = int32_t(fnv1a4(bytes[p], bytes[p + 1], bytes[p + 2], bytes[p + 3]) &
                     uint32_t(vocab_mask));
  }
  __syncthreads();
  n_tok = n_tok_sh;

  // each thread owns dims [4*tid, 4*tid+4) per 1024-dim chunk
  const int dims_per_thread = dim / (ENC_THREADS * 4) > 0 ? dim / ENC_THREADS : 4;
  (void)dims_per_thread;
  // assume dim % (ENC_THREADS*4) == 0 chunks; typical dim=1024 -> 1 chunk of 4
  for (int chunk = 0; chunk < dim; chunk += ENC_THREADS * 4) {
    int d0 = chunk + threadIdx.x * 4;
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
    if (d0 < dim) {
      for (int t = 0; t < n_tok; ++t) {
        const bf16x4 v = *(const bf16x4*)(embed + (size_t)tok[t] * dim + d0);
#pragma unroll
        for (int k = 0; k < 4; ++k) acc[k] += float(v[k]);
      }
    }
    float inv = n_tok > 0 ? 1.0f / float(n_tok) : 0.f;
#pragma unroll
    for (int k = 0; k < 4; ++k) acc[k] *= inv;

    if (normalize) {
      // block reduce sum of squares
      float ss = acc[0] * acc[0] + acc[1] * acc[1] + acc[2] * acc[2] + acc[3] * acc[3];
#pragma unroll
      for (int off = 32; off; off >>= 1) ss += __shfl_down(ss, off);
      __shared__ float wave_ss[ENC_THREADS / WAVE];
      if (lane_id() == 0) wave_ss[wave_id()] = ss;
      __syncthreads();
      if (threadIdx.x == 0) {
        float tot = 0.f;
#pragma unroll
        for (int w = 0; w < ENC_THREADS / WAVE; ++w) tot += wave_ss[w];
        norm_sh = rsqrtf(fmaxf(tot, 1e-12f));
      }
      __syncthreads();
      float r = norm_sh;
#pragma unroll
      for (int k = 0; k < 4; ++k) acc[k] *= r;
      __syncthreads();  // wave_ss reused next chunk
    }

    if (d0 < dim) {
      bf16x4 o;
#pragma unroll
      for (int k = 0; k < 4; ++k) o[k] = bf16(acc[k]);
      *(bf16x4*)(out + (size_t)mi * dim + d0) = o;
    }
  }
}
