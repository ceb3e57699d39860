// philox4x32-10 counter-based RNG — device-side, gfx950.
//
// Bit-identical to the numpy implementation in ops/reference.py
// (the CPU oracle): counter i expands to the 128-bit counter
// (lo32(i), hi32(i), 0, 0); key = (seed_lo, seed_hi or 0x1F123BB5).
// Replaces the reference's numpy RNG draws (stage_3:39-40) on-GPU.
#pragma once
#include <hip/hip_runtime.h>

#define PHILOX_M0 0xD2511F53u
#define PHILOX_M1 0xCD9E8D57u
#define PHILOX_W0 0x9E3779B9u
#define PHILOX_W1 0xBB67AE85u

struct Philox4 {
  unsigned int x, y, z, w;
};

__device__ __forceinline__ Philox4 philox4x32(unsigned long long counter,
                                              unsigned int key0,
                                              unsigned int key1) {
  unsigned int c0 = (unsigned int)(counter & 0xFFFFFFFFull);
  unsigned int c1 = (unsigned int)(counter >> 32);
  unsigned int c2 = 0u, c3 = 0u;
  unsigned int k0 = key0, k1 = key1;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    unsigned long long p0 = (unsigned long long)PHILOX_M0 * c0;
    unsigned long long p1 = (unsigned long long)PHILOX_M1 * c2;
    unsigned int hi0 = (unsigned int)(p0 >> 32), lo0 = (unsigned int)p0;
    unsigned int hi1 = (unsigned int)(p1 >> 32), lo1 = (unsigned int)p1;
    unsigned int n0 = hi1 ^ c1 ^ k0;
    unsigned int n1 = lo1;
    unsigned int n2 = hi0 ^ c3 ^ k1;
    unsigned int n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += PHILOX_W0;
    k1 += PHILOX_W1;
  }
  return {c0, c1, c2, c3};
}

// uniform in [0,1): u32 * 2^-32 (matches the numpy oracle's float32 math)
__device__ __forceinline__ float u32_to_uniform(unsigned int r) {
  return (float)r * 2.3283064365386963e-10f;  // 1/2^32
}
