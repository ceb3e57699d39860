// bf16 vector helpers for gfx950 kernels.
#pragma once
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

typedef __hip_bfloat16 bf16_t;

// 8 bf16 = 16 bytes = one dwordx4 load/store
struct alignas(16) bf16x8 {
  __hip_bfloat162 h[4];
};

__device__ __forceinline__ bf16x8 load_bf16x8(const bf16_t* p) {
  return *(const bf16x8*)p;
}

__device__ __forceinline__ void store_bf16x8(bf16_t* p, const bf16x8& v) {
  *(bf16x8*)p = v;
}

// unpack to 8 floats
__device__ __forceinline__ void bf16x8_to_f32(const bf16x8& v, float* out) {
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    float2 f = __bfloat1622float2(v.h[i]);
    out[2 * i] = f.x;
    out[2 * i + 1] = f.y;
  }
}

__device__ __forceinline__ bf16x8 f32_to_bf16x8(const float* in) {
  bf16x8 v;
#pragma unroll
  for (int i = 0; i < 4; ++i)
    v.h[i] = __float22bfloat162_rn({in[2 * i], in[2 * i + 1]});
  return v;
}

// MFMA operand fragment: 8 bf16 in one 16-B vector
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;

union lds_vec {
  bf16x8_v v;
  short s[8];
};

__device__ __forceinline__ float bf16_to_f32(bf16_t v) {
  return __bfloat162float(v);
}

__device__ __forceinline__ bf16_t f32_to_bf16(float v) {
  return __float2bfloat16(v);
}
