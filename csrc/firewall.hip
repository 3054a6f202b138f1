// Fused per-message firewall tail: verdict aggregation, risk scoring,
// trust updates, audit-record packing.
//
// Implements the reference governance math batched on-GPU:
// - risk factors (risk-assessor.ts:62-99): tool_sensitivity 30,
//   time_of_day 15, trust_deficit 20, frequency 15, target_scope 20
// - verdict precedence deny > 2fa > audit > allow
//   (policy-evaluator.ts:44-78) driven by the DFA hit masks + classifier
//   logits instead of per-policy regex loops
// - trust learning (trust-manager.ts:30-43): violation/success counters
//   updated with atomics, then score = clamp(min(age*.5,20) +
//   min(succ*.1,30) - 2*viol + min(streak*.3,20) + adj, 0, 100)
// - audit record packing: fixed 64-byte binary records hashed by
//   csrc/sha256_merkle.hip into the per-batch Merkle root.
#include "common.hpp"

#define V_ALLOW 0
#define V_AUDIT 1
#define V_2FA 2
#define V_DENY 3

struct __align__(8) AuditRecord64 {
  uint64_t msg_id;       // 8
  uint64_t inj_hits;     // 16
  uint64_t red_hits;     // 24
  float risk;            // 28
  float trust;           // 32
  int32_t agent;         // 36
  uint8_t verdict;       // 37
  uint8_t reserved[3];   // 40
  int64_t ts_ms;         // 48
  float inj_score;       // 52
  uint32_t batch_seq;    // 56
  uint64_t pad;          // 64
};
static_assert(sizeof(AuditRecord64) == 64, "audit record must be 64 B");

extern "C" __global__ void firewall_verdict_kernel(
    const unsigned long long* __restrict__ inj_hits,   // [B]
    const unsigned long long* __restrict__ red_hits,   // [B]
    const float* __restrict__ logits,                  // [B, n_cls] sigmoid'd
    int n_cls,
    const int32_t* __restrict__ agent_idx,             // [B]
    const float* __restrict__ agent_trust,             // [A] current scores
    const float* __restrict__ tool_risk,               // [B] 0..100 per message
    const int32_t* __restrict__ freq_count,            // [B] 60s window count
    int hour, unsigned long long cred_bits, float inj_threshold,
    int8_t* __restrict__ verdict, float* __restrict__ risk,
    float* __restrict__ success_delta,                 // [A] out (atomic)
    float* __restrict__ violation_delta,               // [A] out (atomic)
    int B) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  int a = agent_idx[i];
  float trust = agent_trust[a];

  // risk factors (weights exactly as risk-assessor.ts)
  float f_tool = tool_risk[i] * 0.01f * 30.0f;
  float f_time = (hour < 8 || hour >= 23) ? 15.0f : 0.0f;
  float f_trust = (100.0f - trust) * 0.01f * 20.0f;
  float f_freq = fminf(float(freq_count[i]) / 20.0f, 1.0f) * 15.0f;
  unsigned long long inj = inj_hits[i];
  unsigned long long red = red_hits[i];
  float inj_score = 0.0f;
  for (int c = 0; c < n_cls; ++c) inj_score = fmaxf(inj_score, logits[(size_t)i * n_cls + c]);
  float f_scope = (inj != 0ull || inj_score > inj_threshold) ? 20.0f : 0.0f;
  float r = fminf(f_tool + f_time + f_trust + f_freq + f_scope, 100.0f);
  risk[i] = r;

  // verdict: credential hit or injection pattern -> deny; high classifier
  // score with low trust -> deny; moderate risk -> 2fa; risky-but-logged ->
  // audit; else allow. Precedence deny > 2fa > audit > allow.
  int v = V_ALLOW;
  bool cred = (red & cred_bits) != 0ull;
  bool injected = inj != 0ull || inj_score > inj_threshold;
  if (cred || (injected && trust < 60.0f)) {
    v = V_DENY;
  } else if (injected || (r > 75.0f && trust < 80.0f)) {
    v = V_2FA;
  } else if (r > 50.0f) {
    v = V_AUDIT;
  }
  verdict[i] = (int8_t)v;

  // trust learning: deny counts a violation, allow counts a success
  if (v == V_DENY) {
    atomicAdd(&violation_delta[a], 1.0f);
  } else {
    atomicAdd(&success_delta[a], 1.0f);
  }
}

// Recompute agent scores from accumulated signals (trust-manager.ts:30-43).
extern "C" __global__ void trust_recompute_kernel(
    float* __restrict__ success_count, float* __restrict__ violation_count,
    const float* __restrict__ success_delta, const float* __restrict__ violation_delta,
    const float* __restrict__ age_days, float* __restrict__ clean_streak,
    const float* __restrict__ manual_adj, float* __restrict__ score, int A) {
  int a = blockIdx.x * blockDim.x + threadIdx.x;
  if (a >= A) return;
  float sc = success_count[a] + success_delta[a];
  float vc = violation_count[a] + violation_delta[a];
  success_count[a] = sc;
  violation_count[a] = vc;
  float streak = violation_delta[a] > 0.0f ? 0.0f : clean_streak[a];
  clean_streak[a] = streak;
  float s = fminf(age_days[a] * 0.5f, 20.0f) + fminf(sc * 0.1f, 30.0f) -
            2.0f * vc + fminf(streak * 0.3f, 20.0f) + manual_adj[a];
  score[a] = fminf(fmaxf(s, 0.0f), 100.0f);
}

// Pack fixed 64-byte audit records for the Merkle kernel.
extern "C" __global__ void audit_pack_kernel(
    const int8_t* __restrict__ verdict, const float* __restrict__ risk,
    const unsigned long long* __restrict__ inj_hits,
    const unsigned long long* __restrict__ red_hits,
    const int32_t* __restrict__ agent_idx, const float* __restrict__ agent_trust,
    const float* __restrict__ inj_score, long long ts_ms, long long msg_id0,
    uint32_t batch_seq, AuditRecord64* __restrict__ out, int B) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  AuditRecord64 r;
  r.msg_id = (uint64_t)(msg_id0 + i);
  r.inj_hits = inj_hits[i];
  r.red_hits = red_hits[i];
  r.risk = risk[i];
  int a = agent_idx[i];
  r.trust = agent_trust[a];
  r.agent = a;
  r.verdict = (uint8_t)verdict[i];
  r.reserved[0] = r.reserved[1] = r.reserved[2] = 0;
  r.ts_ms = ts_ms;
  r.inj_score = inj_score ? inj_score[i] : 0.0f;
  r.batch_seq = batch_seq;
  r.pad = 0ull;
  out[i] = r;
}
