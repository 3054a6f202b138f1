// Empirically map the fp4 (blgp=4) B-operand layout of
// v_mfma_scale_f32_16x16x128_f8f6f4.
//
// A is fp8 bit-plane rows: A[r][k] = bit r of k (r<7), A[7][k] = 1.
// Each launch sets exactly ONE nibble (code 2 = +1.0) at byte `byte_pos`,
// nibble `nib`, in the B fragment of lanes with kgrp == `kg`. Reading
// D[r][col] / D[7][col] recovers which k that nibble fed. A second pass
// fixes the position and sweeps all 16 codes to dump the decode table.
#include <hip/hip_runtime.h>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef int v8i __attribute__((ext_vector_type(8)));
typedef int v4i __attribute__((ext_vector_type(4)));

#define HIP_CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  fprintf(stderr, "HIP err %s @%d\n", hipGetErrorString(e), __LINE__); exit(2);} } while (0)

__global__ void map_kernel(const uint8_t* A, float* D, int byte_pos, int nib,
                           int kg_target, int code) {
  int lane = threadIdx.x & 63;
  int row = lane & 15;
  int kg = lane >> 4;
  v8i av = *(const v8i*)(A + row * 128 + kg * 32);
  uint8_t bbytes[16] = {};
  if (kg == kg_target)
    bbytes[byte_pos] = (uint8_t)(nib ? (code << 4) : code);
  v4i b4 = *(const v4i*)bbytes;
  v8i bv = {b4.x, b4.y, b4.z, b4.w, 0, 0, 0, 0};
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      av, bv, c, 0, 4, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[(kg * 4 + r) * 16 + (lane & 15)] = c[r];
}

// all nibbles of kg_target = +1.0; scale_b has byte `sbyte` = 0x80 (x2):
// D bit-plane rows reveal which k positions the byte scales
__global__ void scale_map_kernel(const uint8_t* A, float* D, int kg_target,
                                 int sbyte) {
  int lane = threadIdx.x & 63;
  int row = lane & 15;
  int kg = lane >> 4;
  v8i av = *(const v8i*)(A + row * 128 + kg * 32);
  uint8_t bbytes[16];
#pragma unroll
  for (int i = 0; i < 16; ++i) bbytes[i] = (kg == kg_target) ? 0x22 : 0;
  v4i b4 = *(const v4i*)bbytes;
  v8i bv = {b4.x, b4.y, b4.z, b4.w, 0, 0, 0, 0};
  int sb = (0x7F7F7F7F & ~(0xFF << (8 * sbyte))) | (0x80 << (8 * sbyte));
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      av, bv, c, 0, 4, 0, 0x7F7F7F7F, 0, sb);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[(kg * 4 + r) * 16 + (lane & 15)] = c[r];
}

static uint8_t fp8_one = 0x38;  // e4m3 1.0

// A-side fp4 SCALE semantics: A all-ones fp4 in kg_target, scale_a byte
// sbyte = 0x80 (x2), B fp8 bit-plane cols: bit sums show which k doubled
__global__ void scale_a_map_kernel(const uint8_t* B, float* D, int kg_target,
                                   int sbyte) {
  int lane = threadIdx.x & 63;
  int row = lane & 15;
  int kg = lane >> 4;
  uint8_t abytes[16];
#pragma unroll
  for (int i = 0; i < 16; ++i) abytes[i] = (kg == kg_target) ? 0x22 : 0;
  v4i a4 = *(const v4i*)abytes;
  v8i av = {a4.x, a4.y, a4.z, a4.w, 0, 0, 0, 0};
  v8i bv = *(const v8i*)(B + row * 128 + kg * 32);
  int sa = (0x7F7F7F7F & ~(0xFF << (8 * sbyte))) | (0x80 << (8 * sbyte));
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      av, bv, c, 4, 0, 0, sa, 0, 0x7F7F7F7F);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[(kg * 4 + r) * 16 + (lane & 15)] = c[r];
}

// A-side fp4 map: A one-hot fp4 nibble (cbsz=4), B fp8 bit-plane COLS
// (B[col][k] = bit col of k). D[0][c]/D[0][7] recovers the fed k.
__global__ void map_a_kernel(const uint8_t* B, float* D, int byte_pos, int nib,
                             int kg_target, int code) {
  int lane = threadIdx.x & 63;
  int row = lane & 15;
  int kg = lane >> 4;
  uint8_t abytes[16] = {};
  if (kg == kg_target)
    abytes[byte_pos] = (uint8_t)(nib ? (code << 4) : code);
  v4i a4 = *(const v4i*)abytes;
  v8i av = {a4.x, a4.y, a4.z, a4.w, 0, 0, 0, 0};
  v8i bv = *(const v8i*)(B + row * 128 + kg * 32);
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      av, bv, c, 4 /*A fp4*/, 0 /*B fp8*/, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[(kg * 4 + r) * 16 + (lane & 15)] = c[r];
}

int main() {
  const int M = 16, K = 128;
  std::vector<uint8_t> ha(M * K, 0);
  for (int k = 0; k < K; ++k) {
    for (int r = 0; r < 7; ++r)
      if ((k >> r) & 1) ha[r * K + k] = fp8_one;
    ha[7 * K + k] = fp8_one;
  }
  uint8_t* da; float* dd;
  HIP_CHECK(hipMalloc(&da, M * K));
  HIP_CHECK(hipMalloc(&dd, M * M * sizeof(float)));
  HIP_CHECK(hipMemcpy(da, ha.data(), M * K, hipMemcpyHostToDevice));
  std::vector<float> hd(M * M);

  printf("== position map (code=2 => +1.0), kgrp 0 and 1 ==\n");
  for (int kg = 0; kg < 2; ++kg)
    for (int nib = 0; nib < 2; ++nib)
      for (int bp = 0; bp < 16; ++bp) {
        HIP_CHECK(hipMemset(dd, 0, M * M * sizeof(float)));
        hipLaunchKernelGGL(map_kernel, dim3(1), dim3(64), 0, 0, da, dd, bp, nib, kg, 2);
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipMemcpy(hd.data(), dd, M * M * sizeof(float), hipMemcpyDeviceToHost));
        float mag = hd[7 * M + 0];
        int kk = -1;
        if (fabsf(mag) > 1e-6) {
          kk = 0;
          for (int r = 0; r < 7; ++r)
            if (fabsf(hd[r * M + 0] / mag - 1.0f) < 0.25f) kk |= 1 << r;
        }
        printf("kg%d byte%02d nib%d -> k=%3d mag=%.3f\n", kg, bp, nib, kk, mag);
      }

  printf("== A-side fp4 position map (kg0/kg1, nib0/1, bytes 0/1/7/8) ==\n");
  for (int kg = 0; kg < 2; ++kg)
    for (int nib = 0; nib < 2; ++nib)
      for (int bp : {0, 1, 7, 8}) {
        HIP_CHECK(hipMemset(dd, 0, M * M * sizeof(float)));
        hipLaunchKernelGGL(map_a_kernel, dim3(1), dim3(64), 0, 0, da, dd, bp, nib, kg, 2);
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipMemcpy(hd.data(), dd, M * M * sizeof(float), hipMemcpyDeviceToHost));
        float mag = hd[0 * M + 7];  // row 0, col 7 (all-ones B col)
        int kk = -1;
        if (fabsf(mag) > 1e-6) {
          kk = 0;
          for (int r = 0; r < 7; ++r)
            if (fabsf(hd[0 * M + r] / mag - 1.0f) < 0.25f) kk |= 1 << r;
        }
        printf("A kg%d byte%02d nib%d -> k=%3d mag=%.3f\n", kg, bp, nib, kk, mag);
      }

  printf("== A-side scale-byte map (x2 on byte sb; row0 bit-plane cols) ==\n");
  for (int kg = 0; kg < 2; ++kg)
    for (int sbk = 0; sbk < 4; ++sbk) {
      HIP_CHECK(hipMemset(dd, 0, M * M * sizeof(float)));
      hipLaunchKernelGGL(scale_a_map_kernel, dim3(1), dim3(64), 0, 0, da, dd, kg, sbk);
      HIP_CHECK(hipDeviceSynchronize());
      HIP_CHECK(hipMemcpy(hd.data(), dd, M * M * sizeof(float), hipMemcpyDeviceToHost));
      printf("A kg%d sb%d: tot=%5.1f bits:", kg, sbk, hd[0 * M + 7]);
      for (int r = 0; r < 7; ++r) printf(" %5.1f", hd[0 * M + r]);
      printf("\n");
    }

  printf("== scale-byte map: all-ones data in kg, scale byte sb = x2 ==\n");
  printf("   (for each (kg, sb): bit-plane sums s[r] = sum scale(k)*bit_r(k))\n");
  for (int kg = 0; kg < 4; ++kg)
    for (int sb = 0; sb < 4; ++sb) {
      HIP_CHECK(hipMemset(dd, 0, M * M * sizeof(float)));
      hipLaunchKernelGGL(scale_map_kernel, dim3(1), dim3(64), 0, 0, da, dd, kg, sb);
      HIP_CHECK(hipDeviceSynchronize());
      HIP_CHECK(hipMemcpy(hd.data(), dd, M * M * sizeof(float), hipMemcpyDeviceToHost));
      printf("kg%d sb%d: tot=%5.1f bits:", kg, sb, hd[7 * M + 0]);
      for (int r = 0; r < 7; ++r) printf(" %5.1f", hd[r * M + 0]);
      printf("\n");
    }

  printf("== decode table (kg0 byte0 nib0, codes 0..15) ==\n");
  for (int code = 0; code < 16; ++code) {
    HIP_CHECK(hipMemset(dd, 0, M * M * sizeof(float)));
    hipLaunchKernelGGL(map_kernel, dim3(1), dim3(64), 0, 0, da, dd, 0, 0, 0, code);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(hd.data(), dd, M * M * sizeof(float), hipMemcpyDeviceToHost));
    printf("code %2d -> %.4f\n", code, hd[7 * M + 0]);
  }
  return 0;
}
