// Common device helpers for the cyclegan_amd gfx950 kernels.
// CDNA4: wave64, MFMA bf16 16x16x32, LDS 160KiB/CU. Compiled only for
// --offload-arch=gfx950 (no multi-arch dispatch).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEV __device__ __forceinline__

typedef __bf16 bf16r;
typedef bf16r v8bf __attribute__((ext_vector_type(8)));
typedef float v4f __attribute__((ext_vector_type(4)));
typedef short v8s __attribute__((ext_vector_type(8)));

// raw bf16 <-> f32 (RNE via hardware conversion)
DEV float b2f(short raw) {
  union { float f; unsigned u; } v;
  v.u = ((unsigned)(unsigned short)raw) << 16;
  return v.f;
}

DEV short f2b(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);  // RNE
  return *reinterpret_cast<short*>(&h);
}

// activation codes shared with python (ops/conv.py)
#define ACT_NONE 0
#define ACT_RELU 1
#define ACT_LRELU 2
#define ACT_TANH 3

DEV float apply_act(float v, int act, float slope) {
  switch (act) {
    case ACT_RELU: return v > 0.f ? v : 0.f;
    case ACT_LRELU: return v > 0.f ? v : v * slope;
    case ACT_TANH: return tanhf(v);
    default: return v;
  }
}

// reflect index into [0, n) for single reflection (pad < n)
DEV int mirror_idx(int i, int n) {
  if (i < 0) i = -i;
  if (i >= n) i = 2 * n - 2 - i;
  return i;
}

#define CHECK_HIP(x) do { hipError_t e = (x); \
  if (e != hipSuccess) { \
    printf("HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__, __LINE__); \
  } } while (0)
