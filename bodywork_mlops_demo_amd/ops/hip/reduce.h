// Wave-64 / block reduction helpers for gfx950 (CDNA4: 64-lane waves).
#pragma once
#include <hip/hip_runtime.h>

// full-wave (64-lane) sum via xor shuffles
__device__ __forceinline__ double wave_sum_f64(double v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_sum_f32(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ double wave_max_f64(double v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmax(v, __shfl_xor(v, off, 64));
  return v;
}

// atomic max for NON-NEGATIVE doubles: the IEEE bit pattern of x>=0 is
// monotonic as an unsigned integer, so u64 atomicMax is a double max.
__device__ __forceinline__ void atomic_max_nonneg_f64(double* addr, double v) {
  atomicMax((unsigned long long*)addr, __double_as_longlong(v));
}

// block-level sum: per-wave shuffle reduce -> LDS -> wave-0 combine,
// then ONE atomicAdd per block (Guideline 12).
// NWAVES = blockDim.x / 64 (<= 16).
template <int NWAVES>
__device__ __forceinline__ double block_sum_f64(double v, double* lds_scratch) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  v = wave_sum_f64(v);
  if (lane == 0) lds_scratch[wave] = v;
  __syncthreads();
  double total = 0.0;
  if (wave == 0) {
    total = (lane < NWAVES) ? lds_scratch[lane] : 0.0;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) total += __shfl_xor(total, off, 64);
  }
  __syncthreads();  // scratch reusable by the caller afterwards
  return total;  // valid in wave 0 (all lanes)
}
