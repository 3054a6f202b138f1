// Shared helpers for the MI355X (gfx950 / CDNA4) kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64
#define DEVINL __device__ __forceinline__

typedef __bf16 bf16;
typedef __bf16 bf16x2 __attribute__((ext_vector_type(2)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

DEVINL uint32_t lane_id() { return threadIdx.x & (WAVE - 1); }
DEVINL uint32_t wave_id() { return threadIdx.x / WAVE; }

#define HIP_CHECK(cmd)                                                        \
  do {                                                                        \
    hipError_t e = (cmd);                                                     \
    if (e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                   \
                               hipGetErrorString(e) + " at " + __FILE__ +     \
                               ":" + std::to_string(__LINE__));               \
    }                                                                         \
  } while (0)
