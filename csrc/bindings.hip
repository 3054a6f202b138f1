// Torch extension bindings for the MI355X kernels.
//
// All entry points take torch tensors on the current CUDA(HIP) device and
// launch on the current stream. u64 masks travel as int64 tensors
// (bitwise-identical).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <stdexcept>
#include <vector>

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

extern "C" {
__global__ void sha256_leaves_kernel(const uint8_t*, const int32_t*, uint8_t*, int);
__global__ void merkle_level_kernel(const uint8_t*, uint8_t*, int);
__global__ void dfa_scan_kernel(const uint8_t*, const int32_t*, const uint16_t*,
                                const unsigned long long*, const unsigned long long*,
                                const uint8_t*, const int32_t*, int,
                                unsigned long long*, int);
__global__ void encode_messages_kernel(const uint8_t*, const int32_t*, const __bf16*,
                                       int, int, __bf16*, int, int);
__global__ void gemm_nt_bf16_kernel(const __bf16*, const __bf16*, float*, __bf16*,
                                    const float*, int, int, int, int, int);
__global__ void topk_recall_kernel(const __bf16*, const __bf16*, int, int, int, int,
                                   int, float*, int32_t*, const float*, int32_t*, int);
__global__ void topk_recall_fp8_kernel(const uint8_t*, const uint8_t*, int, int, int,
                                       int, int, float*, int32_t*, const float*,
                                       int32_t*, int);
__global__ void topk_scan_mx_kernel(const uint8_t*, const uint8_t*, int, int, int,
                                    int, int, float*, int32_t*, const float*,
                                    int32_t*, int);
__global__ void topk_scan_mx4_kernel(const uint8_t*, const uint8_t*, const uint8_t*,
                                     int, int, int, int, int, float*, int32_t*,
                                     const float*, int32_t*, int);
__global__ void topk_scan_fp4_kernel(const uint8_t*, const uint8_t*, const uint8_t*,
                                     const uint8_t*, int, int, int, int, int,
                                     float*, int32_t*, const float*, int32_t*, int);
__global__ void topk_scan_fp4_v2_kernel(const uint8_t*, const uint8_t*, const uint8_t*,
                                        const uint8_t*, int, int, int, int, int,
                                        float*, int32_t*, const float*, int32_t*, int);
__global__ void topk_scan_fp4_v3_kernel(const uint8_t*, const uint8_t*, const uint8_t*,
                                        const uint8_t*, int, int, int, int, int,
                                        float*, int32_t*, const float*, int32_t*, int);
__global__ void topk_scan_fp4_v5_kernel(const uint8_t*, const uint8_t*, const uint8_t*,
                                        const uint8_t*, int, int, int, int, int,
                                        float*, int32_t*, const float*, int32_t*, int);


__global__ void topk_merge_kernel(const float*, const int32_t*, int, int, int,
                                  float*, int32_t*);
__global__ void firewall_verdict_kernel(const unsigned long long*, const unsigned long long*,
                                        const float*, int, const int32_t*, const float*,
                                        const float*, const int32_t*, int,
                                        unsigned long long, float, int8_t*, float*,
                                        float*, float*, int);
__global__ void trust_recompute_kernel(float*, float*, const float*, const float*,
                                       const float*, float*, const float*, float*, int);
__global__ void edit_distance_kernel(const uint8_t*, const int32_t*, int32_t*, int);
__global__ void dfa_scan_multi_kernel(const uint8_t*, const int32_t*, const uint16_t*,
                                      const unsigned long long*, const unsigned long long*,
                                      const uint8_t*, const int32_t*, const int32_t*,
                                      int, unsigned long long*, int, int);
__global__ void fact_probe_kernel(const uint8_t*, const int32_t*,
                                  const unsigned long long*, const unsigned long long*,
                                  const unsigned long long*, const unsigned long long*,
                                  int, int32_t*, int32_t*, int);
struct AuditRecord64;
__global__ void audit_pack_kernel(const int8_t*, const float*, const unsigned long long*,
                                  const unsigned long long*, const int32_t*, const float*,
                                  const float*, long long, long long, uint32_t,
                                  AuditRecord64*, int);
}

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

torch::Tensor sha256_leaves(torch::Tensor bytes, torch::Tensor offsets) {
  CHECK_GPU(bytes); CHECK_CONTIG(bytes); CHECK_GPU(offsets); CHECK_CONTIG(offsets);
  TORCH_CHECK(bytes.dtype() == torch::kUInt8 && offsets.dtype() == torch::kInt32);
  int n = offsets.numel() - 1;
  auto out = torch::empty({n, 32}, bytes.options());
  int threads = 128;
  int blocks = (n + threads - 1) / threads;
  hipLaunchKernelGGL(sha256_leaves_kernel, dim3(blocks), dim3(threads), 0, cur_stream(),
                     bytes.data_ptr<uint8_t>(), offsets.data_ptr<int32_t>(),
                     out.data_ptr<uint8_t>(), n);
  return out;
}

torch::Tensor merkle_root_gpu(torch::Tensor digests) {
  CHECK_GPU(digests); CHECK_CONTIG(digests);
  TORCH_CHECK(digests.dtype() == torch::kUInt8 && digests.size(1) == 32);
  int n = digests.size(0);
  TORCH_CHECK(n >= 1);
  auto cur = digests;
  while (n > 1) {
    int n_out = (n + 1) / 2;
    auto nxt = torch::empty({n_out, 32}, digests.options());
    int threads = 128;
    int blocks = (n_out + threads - 1) / threads;
    hipLaunchKernelGGL(merkle_level_kernel, dim3(blocks), dim3(threads), 0, cur_stream(),
                       cur.data_ptr<uint8_t>(), nxt.data_ptr<uint8_t>(), n);
    cur = nxt;
    n = n_out;
  }
  return cur.reshape({32});
}

torch::Tensor dfa_scan(torch::Tensor bytes, torch::Tensor offsets, torch::Tensor next_tab,
                       torch::Tensor accept, torch::Tensor eof_mask,
                       torch::Tensor class_maps, torch::Tensor meta) {
  CHECK_GPU(bytes); CHECK_GPU(offsets); CHECK_GPU(next_tab); CHECK_GPU(accept);
  CHECK_GPU(eof_mask); CHECK_GPU(class_maps); CHECK_GPU(meta);
  TORCH_CHECK(next_tab.dtype() == torch::kInt16 || next_tab.dtype() == torch::kUInt8 ||
              next_tab.scalar_type() == at::kShort, "next_tab must be int16 view of u16");
  int n = offsets.numel() - 1;
  int n_dfas = meta.size(0);
  auto hits = torch::zeros({n}, torch::dtype(torch::kInt64).device(bytes.device()));
  int threads = 256;
  int blocks = (n + threads - 1) / threads;
  size_t lds = (size_t)n_dfas * 256;
  hipLaunchKernelGGL(dfa_scan_kernel, dim3(blocks), dim3(threads), lds, cur_stream(),
                     bytes.data_ptr<uint8_t>(), offsets.data_ptr<int32_t>(),
                     reinterpret_cast<const uint16_t*>(next_tab.data_ptr()),
                     reinterpret_cast<const unsigned long long*>(accept.data_ptr<int64_t>()),
                     reinterpret_cast<const unsigned long long*>(eof_mask.data_ptr<int64_t>()),
                     class_maps.data_ptr<uint8_t>(), meta.data_ptr<int32_t>(), n_dfas,
                     reinterpret_cast<unsigned long long*>(hits.data_ptr<int64_t>()), n);
  return hits;
}

// One launch scanning ALL families: family_ranges [n_families, 2] rows of
// (begin, end) into meta rows; hits come back as int64 [n_families, n].
torch::Tensor dfa_scan_multi(torch::Tensor bytes, torch::Tensor offsets,
                             torch::Tensor next_tab, torch::Tensor accept,
                             torch::Tensor eof_mask, torch::Tensor class_maps,
                             torch::Tensor meta, torch::Tensor family_ranges) {
  CHECK_GPU(bytes); CHECK_GPU(offsets); CHECK_GPU(next_tab); CHECK_GPU(accept);
  CHECK_GPU(eof_mask); CHECK_GPU(class_maps); CHECK_GPU(meta); CHECK_GPU(family_ranges);
  int n = offsets.numel() - 1;
  int n_dfas = meta.size(0);
  int n_families = family_ranges.size(0);
  auto hits = torch::zeros({n_families, n},
                           torch::dtype(torch::kInt64).device(bytes.device()));
  int threads = 256;
  int blocks = (n + threads - 1) / threads;
  size_t lds = (size_t)n_dfas * 256;
  TORCH_CHECK(lds <= 160 * 1024, "class maps exceed LDS");
  hipLaunchKernelGGL(dfa_scan_multi_kernel, dim3(blocks), dim3(threads), lds, cur_stream(),
                     bytes.data_ptr<uint8_t>(), offsets.data_ptr<int32_t>(),
                     reinterpret_cast<const uint16_t*>(next_tab.data_ptr()),
                     reinterpret_cast<const unsigned long long*>(accept.data_ptr<int64_t>()),
                     reinterpret_cast<const unsigned long long*>(eof_mask.data_ptr<int64_t>()),
                     class_maps.data_ptr<uint8_t>(), meta.data_ptr<int32_t>(),
                     family_ranges.data_ptr<int32_t>(), n_families,
                     reinterpret_cast<unsigned long long*>(hits.data_ptr<int64_t>()),
                     n, n_dfas);
  return hits;
}

// GPU fact-registry probe (csrc/fact_probe.hip): returns per-message
// (verified, contradicted) counts for claim-bearing messages.
std::vector<torch::Tensor> fact_probe(torch::Tensor bytes, torch::Tensor offsets,
                                      torch::Tensor claims_mask, torch::Tensor pred_hash,
                                      torch::Tensor table_keys, torch::Tensor table_vals,
                                      int64_t table_pow2) {
  CHECK_GPU(bytes); CHECK_GPU(offsets); CHECK_GPU(claims_mask);
  CHECK_GPU(pred_hash); CHECK_GPU(table_keys); CHECK_GPU(table_vals);
  TORCH_CHECK(pred_hash.numel() == 64, "pred_hash must have 64 entries");
  TORCH_CHECK(table_keys.numel() == (1ll << table_pow2), "table size mismatch");
  int n = offsets.numel() - 1;
  auto opts = torch::dtype(torch::kInt32).device(bytes.device());
  auto verified = torch::zeros({n}, opts);
  auto contradicted = torch::zeros({n}, opts);
  int threads = 256;
  int blocks = (n + threads - 1) / threads;
  hipLaunchKernelGGL(fact_probe_kernel, dim3(blocks), dim3(threads), 0, cur_stream(),
                     bytes.data_ptr<uint8_t>(), offsets.data_ptr<int32_t>(),
                     reinterpret_cast<const unsigned long long*>(claims_mask.data_ptr<int64_t>()),
                     reinterpret_cast<const unsigned long long*>(pred_hash.data_ptr<int64_t>()),
                     reinterpret_cast<const unsigned long long*>(table_keys.data_ptr<int64_t>()),
                     reinterpret_cast<const unsigned long long*>(table_vals.data_ptr<int64_t>()),
                     (int)table_pow2, verified.data_ptr<int32_t>(),
                     contradicted.data_ptr<int32_t>(), n);
  return {verified, contradicted};
}

torch::Tensor encode_messages(torch::Tensor bytes, torch::Tensor offsets,
                              torch::Tensor embed, bool normalize) {
  CHECK_GPU(bytes); CHECK_GPU(offsets); CHECK_GPU(embed); CHECK_CONTIG(embed);
  TORCH_CHECK(embed.dtype() == torch::kBFloat16);
  int vocab = embed.size(0);
  int dim = embed.size(1);
  TORCH_CHECK((vocab & (vocab - 1)) == 0, "vocab must be a power of two");
  TORCH_CHECK(dim % 1024 == 0, "dim must be a multiple of 1024");
  int n = offsets.numel() - 1;
  auto out = torch::empty({n, dim}, embed.options());
  hipLaunchKernelGGL(encode_messages_kernel, dim3(n), dim3(256), 0, cur_stream(),
                     bytes.data_ptr<uint8_t>(), offsets.data_ptr<int32_t>(),
                     reinterpret_cast<const __bf16*>(embed.data_ptr()), vocab - 1, dim,
                     reinterpret_cast<__bf16*>(out.data_ptr()), n, normalize ? 1 : 0);
  return out;
}

torch::Tensor gemm_nt(torch::Tensor A, torch::Tensor B,
                      c10::optional<torch::Tensor> bias, int64_t act, bool out_bf16) {
  CHECK_GPU(A); CHECK_CONTIG(A); CHECK_GPU(B); CHECK_CONTIG(B);
  TORCH_CHECK(A.dtype() == torch::kBFloat16 && B.dtype() == torch::kBFloat16);
  int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "K mismatch");
  TORCH_CHECK(K % 32 == 0, "K must be a multiple of 32");
  auto opts = torch::dtype(out_bf16 ? torch::kBFloat16 : torch::kFloat32).device(A.device());
  auto C = torch::empty({M, N}, opts);
  const float* bias_ptr = nullptr;
  if (bias.has_value()) {
    CHECK_GPU(bias.value());
    TORCH_CHECK(bias->dtype() == torch::kFloat32 && bias->numel() == N);
    bias_ptr = bias->data_ptr<float>();
  }
  dim3 grid((M + 127) / 128, (N + 127) / 128);
  hipLaunchKernelGGL(gemm_nt_bf16_kernel, grid, dim3(256), 0, cur_stream(),
                     reinterpret_cast<const __bf16*>(A.data_ptr()),
                     reinterpret_cast<const __bf16*>(B.data_ptr()),
                     out_bf16 ? nullptr : C.data_ptr<float>(),
                     out_bf16 ? reinterpret_cast<__bf16*>(C.data_ptr()) : nullptr,
                     bias_ptr, M, N, K, (int)act, out_bf16 ? 1 : 0);
  return C;
}

std::vector<torch::Tensor> topk_recall(torch::Tensor Q, torch::Tensor X, int64_t k,
                                       int64_t n_swaths) {
  CHECK_GPU(Q); CHECK_CONTIG(Q); CHECK_GPU(X); CHECK_CONTIG(X);
  TORCH_CHECK(Q.dtype() == torch::kBFloat16 && X.dtype() == torch::kBFloat16);
  int nq = Q.size(0), D = Q.size(1);
  long long nx = X.size(0);
  TORCH_CHECK(X.size(1) == D && D % 64 == 0, "D must be a multiple of 64");
  TORCH_CHECK(k >= 1 && k <= 32, "k in [1,32]");
  int n_qblocks = (nq + 255) / 256;  // BM=256 (topk_recall.hip v3)
  // merge kernel holds n_swaths*k candidates in 16 regs x 64 lanes
  TORCH_CHECK(n_swaths >= 1 && (long long)n_swaths * k <= 1024,
              "n_swaths * k must be <= 1024");
  auto f32opts = torch::dtype(torch::kFloat32).device(Q.device());
  auto i32opts = torch::dtype(torch::kInt32).device(Q.device());
  // candidate state lives in global memory, host-initialized; the kernel
  // only updates slots that beat the running per-row threshold
  auto cand_s = torch::full({(long long)n_qblocks * n_swaths * 256 * k}, -1e30,
                            f32opts);
  auto cand_i = torch::full({(long long)n_qblocks * n_swaths * 256 * k}, -1,
                            i32opts);
  // flat grid: block f -> (qb = f / S, swath = f % S); S a multiple of 8
  // keeps each co-sweeping same-swath group on one XCD
  dim3 grid((unsigned)(n_qblocks * n_swaths));
  // 512 threads = the kernel's 8-wave 2x4 grid (TK_THREADS in topk_recall.hip)
  hipLaunchKernelGGL(topk_recall_kernel, grid, dim3(512), 0, cur_stream(),
                     reinterpret_cast<const __bf16*>(Q.data_ptr()),
                     reinterpret_cast<const __bf16*>(X.data_ptr()), nq, (int)nx, D,
                     (int)k, (int)n_swaths, cand_s.data_ptr<float>(),
                     cand_i.data_ptr<int32_t>(), nullptr, nullptr, 0);
  auto out_s = torch::empty({nq, k}, f32opts);
  auto out_i = torch::empty({nq, k}, i32opts);
  int waves_per_block = 4;
  int blocks = (nq + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(topk_merge_kernel, dim3(blocks), dim3(waves_per_block * 64), 0,
                     cur_stream(), cand_s.data_ptr<float>(), cand_i.data_ptr<int32_t>(),
                     nq, (int)k, (int)n_swaths, out_s.data_ptr<float>(),
                     out_i.data_ptr<int32_t>());
  return {out_s, out_i};
}

std::vector<torch::Tensor> topk_recall_fp8(torch::Tensor Q8, torch::Tensor X8,
                                           int64_t k, int64_t n_swaths) {
  // stage-1 fp8 scan of the two-stage recall: inputs are e4m3 bytes
  // (viewed as uint8), pre-scaled x8; scores rank candidates only.
  CHECK_GPU(Q8); CHECK_CONTIG(Q8); CHECK_GPU(X8); CHECK_CONTIG(X8);
  TORCH_CHECK(Q8.dtype() == torch::kUInt8 && X8.dtype() == torch::kUInt8,
              "fp8 operands passed as uint8 views");
  int nq = Q8.size(0), D = Q8.size(1);
  long long nx = X8.size(0);
  TORCH_CHECK(X8.size(1) == D && D % 64 == 0, "D must be a multiple of 64");
  TORCH_CHECK(k >= 1 && k <= 32, "k in [1,32]");  // TOPK_MAX LDS bound
  int n_qblocks = (nq + 255) / 256;
  TORCH_CHECK(n_swaths >= 1 && (long long)n_swaths * k <= 1024,
              "n_swaths * k must be <= 1024");
  auto f32opts = torch::dtype(torch::kFloat32).device(Q8.device());
  auto i32opts = torch::dtype(torch::kInt32).device(Q8.device());
  auto cand_s = torch::full({(long long)n_qblocks * n_swaths * 256 * k}, -1e30,
                            f32opts);
  auto cand_i = torch::full({(long long)n_qblocks * n_swaths * 256 * k}, -1,
                            i32opts);
  dim3 grid((unsigned)(n_qblocks * n_swaths));
  hipLaunchKernelGGL(topk_recall_fp8_kernel, grid, dim3(512), 0, cur_stream(),
                     Q8.data_ptr<uint8_t>(), X8.data_ptr<uint8_t>(), nq, (int)nx,
                     D, (int)k, (int)n_swaths, cand_s.data_ptr<float>(),
                     cand_i.data_ptr<int32_t>(), nullptr, nullptr, 0);
  auto out_s = torch::empty({nq, k}, f32opts);
  auto out_i = torch::empty({nq, k}, i32opts);
  int waves_per_block = 4;
  int blocks = (nq + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(topk_merge_kernel, dim3(blocks), dim3(waves_per_block * 64), 0,
                     cur_stream(), cand_s.data_ptr<float>(), cand_i.data_ptr<int32_t>(),
                     nq, (int)k, (int)n_swaths, out_s.data_ptr<float>(),
                     out_i.data_ptr<int32_t>());
  return {out_s, out_i};
}

std::vector<torch::Tensor> firewall_verdict(
    torch::Tensor inj_hits, torch::Tensor red_hits, torch::Tensor logits,
    torch::Tensor agent_idx, torch::Tensor agent_trust, torch::Tensor tool_risk,
    torch::Tensor freq_count, int64_t hour, int64_t cred_bits, double inj_threshold,
    int64_t n_agents) {
  CHECK_GPU(inj_hits); CHECK_GPU(red_hits); CHECK_GPU(logits); CHECK_GPU(agent_idx);
  CHECK_GPU(agent_trust); CHECK_GPU(tool_risk); CHECK_GPU(freq_count);
  int B = inj_hits.numel();
  int n_cls = logits.size(1);
  auto verdict = torch::empty({B}, torch::dtype(torch::kInt8).device(inj_hits.device()));
  auto risk = torch::empty({B}, torch::dtype(torch::kFloat32).device(inj_hits.device()));
  auto sdelta = torch::zeros({n_agents}, torch::dtype(torch::kFloat32).device(inj_hits.device()));
  auto vdelta = torch::zeros({n_agents}, torch::dtype(torch::kFloat32).device(inj_hits.device()));
  int threads = 256;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(firewall_verdict_kernel, dim3(blocks), dim3(threads), 0, cur_stream(),
                     reinterpret_cast<const unsigned long long*>(inj_hits.data_ptr<int64_t>()),
                     reinterpret_cast<const unsigned long long*>(red_hits.data_ptr<int64_t>()),
                     logits.data_ptr<float>(), n_cls, agent_idx.data_ptr<int32_t>(),
                     agent_trust.data_ptr<float>(), tool_risk.data_ptr<float>(),
                     freq_count.data_ptr<int32_t>(), (int)hour,
                     (unsigned long long)cred_bits, (float)inj_threshold,
                     verdict.data_ptr<int8_t>(), risk.data_ptr<float>(),
                     sdelta.data_ptr<float>(), vdelta.data_ptr<float>(), B);
  return {verdict, risk, sdelta, vdelta};
}

void trust_recompute(torch::Tensor success_count, torch::Tensor violation_count,
                     torch::Tensor success_delta, torch::Tensor violation_delta,
                     torch::Tensor age_days, torch::Tensor clean_streak,
                     torch::Tensor manual_adj, torch::Tensor score) {
  int A = score.numel();
  int threads = 256;
  int blocks = (A + threads - 1) / threads;
  hipLaunchKernelGGL(trust_recompute_kernel, dim3(blocks), dim3(threads), 0, cur_stream(),
                     success_count.data_ptr<float>(), violation_count.data_ptr<float>(),
                     success_delta.data_ptr<float>(), violation_delta.data_ptr<float>(),
                     age_days.data_ptr<float>(), clean_streak.data_ptr<float>(),
                     manual_adj.data_ptr<float>(), score.data_ptr<float>(), A);
}

torch::Tensor audit_pack(torch::Tensor verdict, torch::Tensor risk, torch::Tensor inj_hits,
                         torch::Tensor red_hits, torch::Tensor agent_idx,
                         torch::Tensor agent_trust, torch::Tensor inj_score,
                         int64_t ts_ms, int64_t msg_id0, int64_t batch_seq) {
  int B = verdict.numel();
  auto out = torch::empty({B, 64}, torch::dtype(torch::kUInt8).device(verdict.device()));
  int threads = 256;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(audit_pack_kernel, dim3(blocks), dim3(threads), 0, cur_stream(),
                     verdict.data_ptr<int8_t>(), risk.data_ptr<float>(),
                     reinterpret_cast<const unsigned long long*>(inj_hits.data_ptr<int64_t>()),
                     reinterpret_cast<const unsigned long long*>(red_hits.data_ptr<int64_t>()),
                     agent_idx.data_ptr<int32_t>(), agent_trust.data_ptr<float>(),
                     inj_score.data_ptr<float>(), (long long)ts_ms, (long long)msg_id0,
                     (uint32_t)batch_seq,
                     reinterpret_cast<AuditRecord64*>(out.data_ptr<uint8_t>()), B);
  return out;
}


std::vector<torch::Tensor> topk_scan_threshold_fp4x4(
    torch::Tensor Q4, torch::Tensor QS, torch::Tensor X4, torch::Tensor XS,
    torch::Tensor theta, int64_t cap, int64_t n_swaths) {
  // both operands MXFP4 (csrc topk_scan_fp4_kernel); scores at x1
  CHECK_GPU(Q4); CHECK_CONTIG(Q4); CHECK_GPU(QS); CHECK_CONTIG(QS);
  CHECK_GPU(X4); CHECK_CONTIG(X4); CHECK_GPU(XS); CHECK_CONTIG(XS);
  CHECK_GPU(theta); CHECK_CONTIG(theta);
  int nq = Q4.size(0), D = (int)Q4.size(1) * 2;
  long long nx = X4.size(0);
  TORCH_CHECK(D % 128 == 0 && D / 32 <= 32, "fp4x4 kernel caps at D=1024");
  TORCH_CHECK(QS.size(0) == nq && QS.size(1) == D / 32);
  TORCH_CHECK(X4.size(1) == D / 2 && XS.size(0) == nx && XS.size(1) == D / 32);
  TORCH_CHECK(theta.numel() == nq && cap >= 32 && cap <= 4096);
  // v3 (2-pair barrier intervals) is the default: bit-exact vs v1 and
  // measured 15.4 vs 16.4 ms at 4.2M x 4096. Env overrides for A/B.
  static const int variant = [] {
    if (const char* e = getenv("VAINPLEX_FP4_V1"); e && e[0] == '1') return 0;
    if (const char* e = getenv("VAINPLEX_FP4_V2"); e && e[0] == '1') return 1;
    if (const char* e = getenv("VAINPLEX_FP4_V5"); e && e[0] == '1') return 4;
    return 2;
  }();
  int n_qblocks = (nq + 255) / 256;
  if (n_swaths <= 0) {
    int want = 256 / (n_qblocks > 0 ? n_qblocks : 1);
    n_swaths = want >= 8 ? (want / 8) * 8 : 8;
    long long max_s = nx / 256; if (max_s < 1) max_s = 1;
    if (n_swaths > max_s) n_swaths = max_s;
  }
  auto f32opts = torch::dtype(torch::kFloat32).device(Q4.device());
  auto i32opts = torch::dtype(torch::kInt32).device(Q4.device());
  auto cand_s = torch::full({(long long)nq, cap}, -1e30, f32opts);
  auto cand_i = torch::full({(long long)nq, cap}, -1, i32opts);
  auto counts = torch::zeros({(long long)nq}, i32opts);
  dim3 grid((unsigned)(n_qblocks * n_swaths));
  auto kern = variant == 4 ? topk_scan_fp4_v5_kernel
            : variant == 2 ? topk_scan_fp4_v3_kernel
            : variant == 1 ? topk_scan_fp4_v2_kernel : topk_scan_fp4_kernel;
  hipLaunchKernelGGL(kern, grid, dim3(512), 0, cur_stream(),
                     Q4.data_ptr<uint8_t>(), QS.data_ptr<uint8_t>(),
                     X4.data_ptr<uint8_t>(), XS.data_ptr<uint8_t>(),
                     nq, (int)nx, D, 1, (int)n_swaths,
                     cand_s.data_ptr<float>(), cand_i.data_ptr<int32_t>(),
                     theta.data_ptr<float>(), counts.data_ptr<int32_t>(),
                     (int)cap);
  return {cand_s, cand_i, counts};
}

std::vector<torch::Tensor> topk_scan_threshold(torch::Tensor Q, torch::Tensor X,
                                               torch::Tensor theta, int64_t cap,
                                               int64_t n_swaths, bool fp8, bool mx) {
  // threshold-scan mode: append every score > theta[q] to a per-query
  // candidate buffer (no top-k maintenance in-kernel)
  CHECK_GPU(Q); CHECK_CONTIG(Q); CHECK_GPU(X); CHECK_CONTIG(X);
  CHECK_GPU(theta); CHECK_CONTIG(theta);
  TORCH_CHECK(theta.dtype() == torch::kFloat32);
  int nq = Q.size(0), D = Q.size(1);
  long long nx = X.size(0);
  TORCH_CHECK(theta.numel() == nq);
  TORCH_CHECK(X.size(1) == D && D % 64 == 0);
  TORCH_CHECK(cap >= 32 && cap <= 4096);
  static const int variant = [] {
    const char* e = getenv("VAINPLEX_FP4_V2");
    return (e != nullptr && e[0] == '1') ? 1 : 0;  // pipelined v2 opt-in
  }();
  int n_qblocks = (nq + 255) / 256;
  if (n_swaths <= 0) {
    int want = 256 / (n_qblocks > 0 ? n_qblocks : 1);
    n_swaths = want >= 8 ? (want / 8) * 8 : 8;
    long long max_s = nx / 256; if (max_s < 1) max_s = 1;
    if (n_swaths > max_s) n_swaths = max_s;
  }
  auto f32opts = torch::dtype(torch::kFloat32).device(Q.device());
  auto i32opts = torch::dtype(torch::kInt32).device(Q.device());
  auto cand_s = torch::full({(long long)nq, cap}, -1e30, f32opts);
  auto cand_i = torch::full({(long long)nq, cap}, -1, i32opts);
  auto counts = torch::zeros({(long long)nq}, i32opts);
  dim3 grid((unsigned)(n_qblocks * n_swaths));
  if (mx) {
    // MX-scaled x128 scan: same e4m3 bytes, unit block scales
    TORCH_CHECK(Q.dtype() == torch::kUInt8 && X.dtype() == torch::kUInt8);
    TORCH_CHECK(D % 128 == 0, "MX scan needs D % 128 == 0");
    hipLaunchKernelGGL(topk_scan_mx_kernel, grid, dim3(512), 0, cur_stream(),
                       Q.data_ptr<uint8_t>(), X.data_ptr<uint8_t>(), nq, (int)nx,
                       D, 1, (int)n_swaths, cand_s.data_ptr<float>(),
                       cand_i.data_ptr<int32_t>(), theta.data_ptr<float>(),
                       counts.data_ptr<int32_t>(), (int)cap);
  } else if (fp8) {
    TORCH_CHECK(Q.dtype() == torch::kUInt8 && X.dtype() == torch::kUInt8);
    hipLaunchKernelGGL(topk_recall_fp8_kernel, grid, dim3(512), 0, cur_stream(),
                       Q.data_ptr<uint8_t>(), X.data_ptr<uint8_t>(), nq, (int)nx,
                       D, 1, (int)n_swaths, cand_s.data_ptr<float>(),
                       cand_i.data_ptr<int32_t>(), theta.data_ptr<float>(),
                       counts.data_ptr<int32_t>(), (int)cap);
  } else {
    TORCH_CHECK(Q.dtype() == torch::kBFloat16 && X.dtype() == torch::kBFloat16);
    hipLaunchKernelGGL(topk_recall_kernel, grid, dim3(512), 0, cur_stream(),
                       reinterpret_cast<const __bf16*>(Q.data_ptr()),
                       reinterpret_cast<const __bf16*>(X.data_ptr()), nq, (int)nx,
                       D, 1, (int)n_swaths, cand_s.data_ptr<float>(),
                       cand_i.data_ptr<int32_t>(), theta.data_ptr<float>(),
                       counts.data_ptr<int32_t>(), (int)cap);
  }
  return {cand_s, cand_i, counts};
}

void register_host_envelope(pybind11::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  register_host_envelope(m);
  m.def("sha256_leaves", &sha256_leaves, "Batched SHA-256 leaf digests");
  m.def("merkle_root", &merkle_root_gpu, "Merkle root over leaf digests");
  m.def("dfa_scan", &dfa_scan, "Multi-pattern DFA scan");
  m.def("dfa_scan_multi", &dfa_scan_multi, "All-family DFA scan in one launch");
  m.def("fact_probe", &fact_probe, "GPU fact-registry hash probe");
  m.def("encode_messages", &encode_messages, "4-gram hash embedding encoder");
  m.def("gemm_nt", &gemm_nt, "bf16 NT GEMM with fused epilogue",
        py::arg("A"), py::arg("B"), py::arg("bias") = c10::nullopt,
        py::arg("act") = 0, py::arg("out_bf16") = false);
  m.def("topk_recall", &topk_recall, "Fused cosine top-k recall");
  m.def("topk_scan_only", [](torch::Tensor Q, torch::Tensor X, int64_t n_swaths, bool fp8,
                             bool mx) {
    // perf diagnosis: run the scan loop with the top-k phase skipped
    CHECK_GPU(Q); CHECK_GPU(X);
    int nq = Q.size(0), D = Q.size(1);
    long long nx = X.size(0);
    int n_qblocks = (nq + 255) / 256;
    auto f32opts = torch::dtype(torch::kFloat32).device(Q.device());
    auto i32opts = torch::dtype(torch::kInt32).device(Q.device());
    auto cand_s = torch::zeros({(long long)n_qblocks * n_swaths * 256}, f32opts);
    auto cand_i = torch::zeros({1}, i32opts);
    dim3 grid((unsigned)(n_qblocks * n_swaths));
    if (mx) {
      hipLaunchKernelGGL(topk_scan_mx_kernel, grid, dim3(512), 0, cur_stream(),
                         Q.data_ptr<uint8_t>(), X.data_ptr<uint8_t>(), nq, (int)nx,
                         D, -1, (int)n_swaths, cand_s.data_ptr<float>(),
                         cand_i.data_ptr<int32_t>(), nullptr, nullptr, 0);
    } else if (fp8) {
      hipLaunchKernelGGL(topk_recall_fp8_kernel, grid, dim3(512), 0, cur_stream(),
                         Q.data_ptr<uint8_t>(), X.data_ptr<uint8_t>(), nq, (int)nx,
                         D, -1, (int)n_swaths, cand_s.data_ptr<float>(),
                         cand_i.data_ptr<int32_t>(), nullptr, nullptr, 0);
    } else {
      hipLaunchKernelGGL(topk_recall_kernel, grid, dim3(512), 0, cur_stream(),
                         reinterpret_cast<const __bf16*>(Q.data_ptr()),
                         reinterpret_cast<const __bf16*>(X.data_ptr()), nq, (int)nx,
                         D, -1, (int)n_swaths, cand_s.data_ptr<float>(),
                         cand_i.data_ptr<int32_t>(), nullptr, nullptr, 0);
    }
  }, "scan-only diagnosis", py::arg("Q"), py::arg("X"), py::arg("n_swaths"),
     py::arg("fp8") = false, py::arg("mx") = false);
  m.def("topk_recall_fp8", &topk_recall_fp8, "fp8 stage-1 scan of two-stage recall");
  m.def("topk_scan_threshold", &topk_scan_threshold,
        "threshold-scan candidate collection (bf16, fp8 or MX-fp8)",
        py::arg("Q"), py::arg("X"), py::arg("theta"), py::arg("cap"),
        py::arg("n_swaths"), py::arg("fp8") = false, py::arg("mx") = false);
  m.def("topk_scan_threshold_fp4",
        [](torch::Tensor Q8, torch::Tensor X4, torch::Tensor XS,
           torch::Tensor theta, int64_t cap, int64_t n_swaths) {
    // MXFP4 X operand: X4 = packed e2m1 nibbles [nx, D/2], XS = e8m0
    // block scales [nx, D/32]; Q8 stays e4m3. Scores are x8 (Q scale only).
    CHECK_GPU(Q8); CHECK_CONTIG(Q8); CHECK_GPU(X4); CHECK_CONTIG(X4);
    CHECK_GPU(XS); CHECK_CONTIG(XS); CHECK_GPU(theta); CHECK_CONTIG(theta);
    TORCH_CHECK(Q8.dtype() == torch::kUInt8 && X4.dtype() == torch::kUInt8 &&
                XS.dtype() == torch::kUInt8);
    TORCH_CHECK(theta.dtype() == torch::kFloat32);
    int nq = Q8.size(0), D = Q8.size(1);
    long long nx = X4.size(0);
    TORCH_CHECK(D % 128 == 0 && D / 32 <= 64, "fp4 scan needs D%128==0, D<=2048");
    TORCH_CHECK(X4.size(1) == D / 2 && XS.size(0) == nx && XS.size(1) == D / 32);
    TORCH_CHECK(theta.numel() == nq);
    TORCH_CHECK(cap >= 32 && cap <= 4096);
    int n_qblocks = (nq + 255) / 256;
    if (n_swaths <= 0) {
      int want = 256 / (n_qblocks > 0 ? n_qblocks : 1);
      n_swaths = want >= 8 ? (want / 8) * 8 : 8;
      long long max_s = nx / 256; if (max_s < 1) max_s = 1;
      if (n_swaths > max_s) n_swaths = max_s;
    }
    auto f32opts = torch::dtype(torch::kFloat32).device(Q8.device());
    auto i32opts = torch::dtype(torch::kInt32).device(Q8.device());
    auto cand_s = torch::full({(long long)nq, cap}, -1e30, f32opts);
    auto cand_i = torch::full({(long long)nq, cap}, -1, i32opts);
    auto counts = torch::zeros({(long long)nq}, i32opts);
    dim3 grid((unsigned)(n_qblocks * n_swaths));
    hipLaunchKernelGGL(topk_scan_mx4_kernel, grid, dim3(512), 0, cur_stream(),
                       Q8.data_ptr<uint8_t>(), X4.data_ptr<uint8_t>(),
                       XS.data_ptr<uint8_t>(), nq, (int)nx, D, 1, (int)n_swaths,
                       cand_s.data_ptr<float>(), cand_i.data_ptr<int32_t>(),
                       theta.data_ptr<float>(), counts.data_ptr<int32_t>(),
                       (int)cap);
    return std::vector<torch::Tensor>{cand_s, cand_i, counts};
  }, "MXFP4-X threshold scan");
  m.def("topk_scan_threshold_fp4x4", &topk_scan_threshold_fp4x4,
        "full-MXFP4 threshold scan (both operands e2m1 + e8m0 group scales)");
  m.def("edit_distance", [](torch::Tensor bytes, torch::Tensor offsets) {
    // batched Levenshtein over string PAIRS: offsets has 2*n+1 entries;
    // pair i = strings (2i, 2i+1) in the packed byte buffer
    CHECK_GPU(bytes); CHECK_CONTIG(bytes); CHECK_GPU(offsets); CHECK_CONTIG(offsets);
    TORCH_CHECK(bytes.dtype() == torch::kUInt8 && offsets.dtype() == torch::kInt32);
    TORCH_CHECK(offsets.numel() % 2 == 1, "offsets must cover 2*n strings");
    int n_pairs = (int)(offsets.numel() / 2);
    auto out = torch::empty({n_pairs}, torch::dtype(torch::kInt32).device(bytes.device()));
    if (n_pairs > 0)
      hipLaunchKernelGGL(edit_distance_kernel, dim3(n_pairs), dim3(64), 0, cur_stream(),
                         bytes.data_ptr<uint8_t>(), offsets.data_ptr<int32_t>(),
                         out.data_ptr<int32_t>(), n_pairs);
    return out;
  }, "Batched Levenshtein edit distance (wavefront DP)");
  m.def("firewall_verdict", &firewall_verdict, "Fused verdict/risk/trust-delta");
  m.def("trust_recompute", &trust_recompute, "Agent trust score recompute");
  m.def("audit_pack", &audit_pack, "Pack 64-byte audit records");
}
