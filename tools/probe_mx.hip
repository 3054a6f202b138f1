// Numeric probe for v_mfma_scale_f32_16x16x128_f8f6f4 on gfx950.
//
// Validates the assumed operand mapping before the MX scan kernel uses it:
//   A (16x128 fp8 e4m3): row = lane&15, k = (lane>>4)*32 + i, i in [0,32)
//   B (16x128 fp8, B^T convention: cols of the 16x16 output): col = lane&15,
//       same k mapping
//   C/D: col = lane&15, row = (lane>>4)*4 + r  (same as every 16x16 shape)
//   scales: e8m0 biased-127 exponent per 32-block; 0x7F == x1.0; probe also
//   checks a non-unit scale byte (0x80 == x2) applied through opsel byte 0.
//
// Usage: probe_mx   (prints PASS/FAIL per check; exit 0 iff all pass)
#include <hip/hip_runtime.h>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef int v8i __attribute__((ext_vector_type(8)));

#define HIP_CHECK(x)                                                        \
  do {                                                                      \
    hipError_t e = (x);                                                     \
    if (e != hipSuccess) {                                                  \
      fprintf(stderr, "HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      exit(2);                                                              \
    }                                                                       \
  } while (0)

// one wave: each lane loads its assumed fragment and runs the MFMA
__global__ void mx_probe_kernel(const uint8_t* A, const uint8_t* B, float* D,
                                int scale_a, int scale_b) {
  int lane = threadIdx.x & 63;
  int row = lane & 15;
  int kg = lane >> 4;
  v8i av, bv;
  const uint8_t* ap = A + row * 128 + kg * 32;
  const uint8_t* bp = B + row * 128 + kg * 32;  // "row" of B == output col
  av = *(const v8i*)ap;
  bv = *(const v8i*)bp;
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      av, bv, c, 0 /*cbsz: A fp8*/, 0 /*blgp: B fp8*/, 0, scale_a, 0, scale_b);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[(kg * 4 + r) * 16 + (lane & 15)] = c[r];
}

// e4m3 OCP encode (round-to-nearest-even on the mantissa, no inf)
static uint8_t f32_to_e4m3(float f) {
  if (f == 0.0f) return 0;
  uint8_t s = f < 0 ? 0x80 : 0;
  float a = fabsf(f);
  int e = (int)floorf(log2f(a));
  if (e < -6) e = -6;
  float m = a / exp2f((float)e);
  int mant = (int)roundf((m - 1.0f) * 8.0f);
  if (mant == 8) { mant = 0; e += 1; }
  if (e > 8) { e = 8; mant = 6; }  // clamp to 448
  int be = e + 7;
  if (m < 1.0f) { be = 0; mant = (int)roundf(a / exp2f(-9.0f)); if (mant > 7) mant = 7; }
  return s | (uint8_t)(be << 3) | (uint8_t)mant;
}

static float e4m3_to_f32(uint8_t v) {
  int s = (v >> 7) & 1, e = (v >> 3) & 15, m = v & 7;
  float f;
  if (e == 0) f = (m / 8.0f) * exp2f(-6.0f);
  else f = (1.0f + m / 8.0f) * exp2f((float)(e - 7));
  return s ? -f : f;
}

int main() {
  const int M = 16, K = 128;
  std::vector<uint8_t> ha(M * K), hb(M * K);
  srand(7);
  for (auto& v : ha) v = f32_to_e4m3(((rand() % 2000) - 1000) / 500.0f);
  for (auto& v : hb) v = f32_to_e4m3(((rand() % 2000) - 1000) / 500.0f);

  // host reference in double from the DECODED fp8 values
  std::vector<double> ref(M * M, 0.0);
  for (int i = 0; i < M; ++i)
    for (int j = 0; j < M; ++j) {
      double s = 0;
      for (int k = 0; k < K; ++k)
        s += (double)e4m3_to_f32(ha[i * K + k]) * (double)e4m3_to_f32(hb[j * K + k]);
      ref[i * M + j] = s;
    }

  uint8_t *da, *db;
  float* dd;
  HIP_CHECK(hipMalloc(&da, M * K));
  HIP_CHECK(hipMalloc(&db, M * K));
  HIP_CHECK(hipMalloc(&dd, M * M * sizeof(float)));
  HIP_CHECK(hipMemcpy(da, ha.data(), M * K, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(db, hb.data(), M * K, hipMemcpyHostToDevice));

  int fails = 0;
  struct Case { const char* name; int sa, sb; double mult; };
  Case cases[] = {
      {"unit scales (0x7f)", 0x7F7F7F7F, 0x7F7F7F7F, 1.0},
      {"A scale x2 (0x80 byte0)", 0x7F7F7F80, 0x7F7F7F7F, 2.0},
      {"B scale x4 (0x81 byte0)", 0x7F7F7F7F, 0x7F7F7F81, 4.0},
  };
  for (auto& cs : cases) {
    HIP_CHECK(hipMemset(dd, 0, M * M * sizeof(float)));
    hipLaunchKernelGGL(mx_probe_kernel, dim3(1), dim3(64), 0, 0, da, db, dd,
                       cs.sa, cs.sb);
    HIP_CHECK(hipDeviceSynchronize());
    std::vector<float> hd(M * M);
    HIP_CHECK(hipMemcpy(hd.data(), dd, M * M * sizeof(float), hipMemcpyDeviceToHost));
    double maxerr = 0;
    for (int i = 0; i < M * M; ++i)
      maxerr = fmax(maxerr, fabs(hd[i] - ref[i] * cs.mult));
    bool ok = maxerr < 1e-2;
    printf("%-28s maxerr=%.6f  %s\n", cs.name, maxerr, ok ? "PASS" : "FAIL");
    if (!ok) {
      ++fails;
      for (int i = 0; i < 4; ++i)
        printf("  d[0][%d]=%f ref=%f\n", i, hd[i], ref[i] * cs.mult);
    }
  }
  return fails ? 1 : 0;
}
