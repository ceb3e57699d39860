// On-GPU synthetic-drift data generator with stable y>=0 stream compaction.
//
// Replaces the reference's numpy generator + pandas cull
// (stage_3_synthetic_data_generation.py:28-43) with three gfx950 kernels:
//   1. generate: philox4x32 -> X ~ U(0,100), eps ~ N(0,1) (Box-Muller),
//      y = alpha + beta*X + sigma*eps; per-block keep-counts (y >= 0).
//   2. two-level exclusive scan of the per-block counts (hand-written,
//      deterministic).
//   3. scatter: stable compaction using wave ballot + popcount offsets.
// Row order is preserved (pandas `query` parity), so the output stream is
// bit-comparable with the CPU oracle.
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include "philox.h"

#define DG_BLOCK 256
#define DG_WAVES (DG_BLOCK / 64)

__global__ void datagen_generate_kernel(
    float* __restrict__ y_tmp, float* __restrict__ X_tmp,
    unsigned int* __restrict__ block_counts, long long n,
    unsigned long long stream_offset, unsigned int key0, unsigned int key1,
    float alpha, float beta, float sigma) {
  const long long i = (long long)blockIdx.x * DG_BLOCK + threadIdx.x;
  int keep = 0;
  if (i < n) {
    Philox4 r = philox4x32(stream_offset + (unsigned long long)i, key0, key1);
    float X = u32_to_uniform(r.x) * 100.0f;
    float u1 = ((float)r.y + 0.5f) * 2.3283064365386963e-10f;
    float u2 = u32_to_uniform(r.z);
    float eps = sqrtf(-2.0f * logf(u1)) * cosf(6.2831853071795864f * u2);
    float y = alpha + beta * X + sigma * eps;
    X_tmp[i] = X;
    y_tmp[i] = y;
    keep = y >= 0.0f;
  }
  // per-block keep count: wave popcount -> LDS -> wave-0 add
  __shared__ unsigned int wave_cnt[DG_WAVES];
  unsigned long long ballot = __ballot(keep);
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) wave_cnt[wave] = (unsigned int)__popcll(ballot);
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned int c = 0;
#pragma unroll
    for (int w = 0; w < DG_WAVES; ++w) c += wave_cnt[w];
    block_counts[blockIdx.x] = c;
  }
}

// ---- two-level exclusive scan over block counts ---------------------------
// level 1: each block scans a 1024-chunk in LDS, writes chunk total
#define SCAN_CHUNK 1024

__global__ void scan_level1_kernel(const unsigned int* __restrict__ in,
                                   unsigned int* __restrict__ out,
                                   unsigned int* __restrict__ chunk_sums,
                                   long long n) {
  __shared__ unsigned int buf[SCAN_CHUNK];
  const long long base = (long long)blockIdx.x * SCAN_CHUNK;
  const int t = threadIdx.x;  // 256 threads, 4 elems each
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    long long idx = base + t + k * 256;
    buf[t + k * 256] = (idx < n) ? in[idx] : 0u;
  }
  __syncthreads();
  // Hillis-Steele inclusive scan in LDS
  for (int off = 1; off < SCAN_CHUNK; off <<= 1) {
    unsigned int vals[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      int idx = t + k * 256;
      vals[k] = (idx >= off) ? buf[idx - off] : 0u;
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < 4; ++k) buf[t + k * 256] += vals[k];
    __syncthreads();
  }
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    long long idx = base + t + k * 256;
    int li = t + k * 256;
    if (idx < n) out[idx] = (li == 0) ? 0u : buf[li - 1];  // exclusive
  }
  if (t == 0) chunk_sums[blockIdx.x] = buf[SCAN_CHUNK - 1];
}

// level 2: single block scans the (<=4096) chunk sums and adds offsets
__global__ void scan_level2_kernel(unsigned int* __restrict__ out,
                                   unsigned int* __restrict__ chunk_sums,
                                   unsigned int* __restrict__ total,
                                   long long n, int n_chunks) {
  // serial-ish scan of chunk sums by thread 0 of wave 0 (n_chunks <= 4096:
  // a few us, off the hot path), then all threads add offsets
  __shared__ unsigned int offs[4096 + 1];
  if (threadIdx.x == 0) {
    unsigned int acc = 0;
    for (int c = 0; c < n_chunks; ++c) {
      offs[c] = acc;
      acc += chunk_sums[c];
    }
    offs[n_chunks] = acc;
    *total = acc;
  }
  __syncthreads();
  const long long stride = (long long)blockDim.x * gridDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    out[i] += offs[i / SCAN_CHUNK];
  }
}

__global__ void datagen_scatter_kernel(
    const float* __restrict__ y_tmp, const float* __restrict__ X_tmp,
    const unsigned int* __restrict__ block_offsets,
    float* __restrict__ y_out, float* __restrict__ X_out, long long n) {
  const long long i = (long long)blockIdx.x * DG_BLOCK + threadIdx.x;
  float y = 0.f, X = 0.f;
  int keep = 0;
  if (i < n) {
    y = y_tmp[i];
    X = X_tmp[i];
    keep = y >= 0.0f;
  }
  __shared__ unsigned int wave_off[DG_WAVES];
  unsigned long long ballot = __ballot(keep);
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) wave_off[wave] = (unsigned int)__popcll(ballot);
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned int acc = 0;
#pragma unroll
    for (int w = 0; w < DG_WAVES; ++w) {
      unsigned int c = wave_off[w];
      wave_off[w] = acc;
      acc += c;
    }
  }
  __syncthreads();
  if (keep) {
    unsigned int before =
        (unsigned int)__popcll(ballot & ((1ull << lane) - 1ull));
    unsigned int pos = block_offsets[blockIdx.x] + wave_off[wave] + before;
    y_out[pos] = y;
    X_out[pos] = X;
  }
}

std::tuple<at::Tensor, at::Tensor> datagen_hip(
    int64_t n, int64_t seed, int64_t stream_offset, double alpha, double beta,
    double sigma) {
  TORCH_CHECK(n > 0, "datagen: n must be positive");
  auto opts = at::TensorOptions().dtype(at::kFloat).device(at::kCUDA);
  auto y_tmp = at::empty({n}, opts);
  auto X_tmp = at::empty({n}, opts);
  const long long n_blocks = (n + DG_BLOCK - 1) / DG_BLOCK;
  const long long n_chunks = (n_blocks + SCAN_CHUNK - 1) / SCAN_CHUNK;
  TORCH_CHECK(n_chunks <= 4096,
              "datagen: n too large for the two-level scan (max ~1.07e9 rows "
              "per call; shard across calls)");
  auto u32 = at::TensorOptions().dtype(at::kUInt32).device(at::kCUDA);
  auto counts = at::empty({n_blocks}, u32);
  auto offsets = at::empty({n_blocks}, u32);
  auto chunk_sums = at::empty({std::max<long long>(n_chunks, 1)}, u32);
  auto total = at::zeros({1}, u32);

  unsigned int key0 = (unsigned int)(seed & 0xFFFFFFFFll);
  unsigned int key1 = (seed > 0xFFFFFFFFll) ? (unsigned int)(seed >> 32)
                                            : 0x1F123BB5u;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(datagen_generate_kernel, dim3(n_blocks), dim3(DG_BLOCK),
                     0, stream,
                     y_tmp.data_ptr<float>(), X_tmp.data_ptr<float>(),
                     (unsigned int*)counts.data_ptr(), (long long)n,
                     (unsigned long long)stream_offset, key0, key1,
                     (float)alpha, (float)beta, (float)sigma);
  hipLaunchKernelGGL(scan_level1_kernel, dim3(n_chunks), dim3(256), 0, stream,
                     (const unsigned int*)counts.data_ptr(),
                     (unsigned int*)offsets.data_ptr(),
                     (unsigned int*)chunk_sums.data_ptr(), (long long)n_blocks);
  hipLaunchKernelGGL(scan_level2_kernel,
                     dim3(std::min<long long>(2048, n_chunks * 4)), dim3(256),
                     0, stream,
                     (unsigned int*)offsets.data_ptr(),
                     (unsigned int*)chunk_sums.data_ptr(),
                     (unsigned int*)total.data_ptr(), (long long)n_blocks,
                     (int)n_chunks);
  auto y_out = at::empty({n}, opts);
  auto X_out = at::empty({n}, opts);
  hipLaunchKernelGGL(datagen_scatter_kernel, dim3(n_blocks), dim3(DG_BLOCK), 0,
                     stream,
                     y_tmp.data_ptr<float>(), X_tmp.data_ptr<float>(),
                     (const unsigned int*)offsets.data_ptr(),
                     y_out.data_ptr<float>(), X_out.data_ptr<float>(),
                     (long long)n);
  // one D2H sync to size the result (datagen is per-cycle, not per-request)
  int64_t kept = total.to(at::kLong).item<int64_t>();
  return {y_out.narrow(0, 0, kept), X_out.narrow(0, 0, kept)};
}

// ---- random 80/20 split (replaces sklearn train_test_split, stage_1:98) ---
// Element i is a TEST row iff philox(seed, i).x < tau; both partitions keep
// row order (stable).  Single pass + the same two-level scan as datagen:
// no host-side permutation, no gather indices — the reference's CPU
// shuffle was 2/3 of the GPU cycle's train phase before this kernel.

__global__ void split_flag_kernel(unsigned int* __restrict__ block_counts,
                                  long long n, unsigned int key0,
                                  unsigned int key1, unsigned int tau) {
  const long long i = (long long)blockIdx.x * DG_BLOCK + threadIdx.x;
  int is_test = 0;
  if (i < n) {
    Philox4 r = philox4x32((unsigned long long)i, key0, key1);
    is_test = r.x < tau;
  }
  __shared__ unsigned int wave_cnt[DG_WAVES];
  unsigned long long ballot = __ballot(is_test);
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) wave_cnt[wave] = (unsigned int)__popcll(ballot);
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned int c = 0;
#pragma unroll
    for (int w = 0; w < DG_WAVES; ++w) c += wave_cnt[w];
    block_counts[blockIdx.x] = c;
  }
}

__global__ void split_scatter_kernel(
    const float* __restrict__ X, const float* __restrict__ y,
    const unsigned int* __restrict__ test_offsets,
    float* __restrict__ X_tr, float* __restrict__ y_tr,
    float* __restrict__ X_te, float* __restrict__ y_te, long long n,
    unsigned int key0, unsigned int key1, unsigned int tau) {
  const long long i = (long long)blockIdx.x * DG_BLOCK + threadIdx.x;
  int is_test = 0;
  if (i < n) {
    Philox4 r = philox4x32((unsigned long long)i, key0, key1);
    is_test = r.x < tau;
  }
  __shared__ unsigned int wave_off[DG_WAVES];
  unsigned long long ballot = __ballot(is_test);
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) wave_off[wave] = (unsigned int)__popcll(ballot);
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned int acc = 0;
#pragma unroll
    for (int w = 0; w < DG_WAVES; ++w) {
      unsigned int c = wave_off[w];
      wave_off[w] = acc;
      acc += c;
    }
  }
  __syncthreads();
  if (i < n) {
    unsigned int before =
        (unsigned int)__popcll(ballot & ((1ull << lane) - 1ull));
    unsigned int tests_before =
        test_offsets[blockIdx.x] + wave_off[wave] + before;
    if (is_test) {
      X_te[tests_before] = X[i];
      y_te[tests_before] = y[i];
    } else {
      long long train_pos = i - (long long)tests_before;
      X_tr[train_pos] = X[i];
      y_tr[train_pos] = y[i];
    }
  }
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> random_split_hip(
    const at::Tensor& X, const at::Tensor& y, double test_frac,
    int64_t seed) {
  TORCH_CHECK(X.is_cuda() && y.is_cuda() && X.numel() == y.numel());
  TORCH_CHECK(X.scalar_type() == at::kFloat && y.scalar_type() == at::kFloat);
  long long n = X.numel();
  const long long n_blocks = (n + DG_BLOCK - 1) / DG_BLOCK;
  const long long n_chunks = (n_blocks + SCAN_CHUNK - 1) / SCAN_CHUNK;
  TORCH_CHECK(n_chunks <= 4096, "random_split: n too large (max ~1.07e9)");
  auto u32 = at::TensorOptions().dtype(at::kUInt32).device(X.device());
  auto counts = at::empty({n_blocks}, u32);
  auto offsets = at::empty({n_blocks}, u32);
  auto chunk_sums = at::empty({std::max<long long>(n_chunks, 1)}, u32);
  auto total = at::zeros({1}, u32);
  unsigned int key0 = (unsigned int)(seed & 0xFFFFFFFFll);
  unsigned int key1 = (seed > 0xFFFFFFFFll) ? (unsigned int)(seed >> 32)
                                            : 0x85EBCA6Bu;  // split stream
  unsigned int tau = (unsigned int)(test_frac * 4294967296.0);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(split_flag_kernel, dim3(n_blocks), dim3(DG_BLOCK), 0,
                     stream, (unsigned int*)counts.data_ptr(), n, key0, key1,
                     tau);
  hipLaunchKernelGGL(scan_level1_kernel, dim3(n_chunks), dim3(256), 0, stream,
                     (const unsigned int*)counts.data_ptr(),
                     (unsigned int*)offsets.data_ptr(),
                     (unsigned int*)chunk_sums.data_ptr(), (long long)n_blocks);
  hipLaunchKernelGGL(scan_level2_kernel,
                     dim3(std::min<long long>(2048, n_chunks * 4)), dim3(256),
                     0, stream, (unsigned int*)offsets.data_ptr(),
                     (unsigned int*)chunk_sums.data_ptr(),
                     (unsigned int*)total.data_ptr(), (long long)n_blocks,
                     (int)n_chunks);
  auto opts = X.options();
  auto X_tr = at::empty({n}, opts);
  auto y_tr = at::empty({n}, opts);
  auto X_te = at::empty({n}, opts);
  auto y_te = at::empty({n}, opts);
  hipLaunchKernelGGL(split_scatter_kernel, dim3(n_blocks), dim3(DG_BLOCK), 0,
                     stream, X.data_ptr<float>(), y.data_ptr<float>(),
                     (const unsigned int*)offsets.data_ptr(),
                     X_tr.data_ptr<float>(), y_tr.data_ptr<float>(),
                     X_te.data_ptr<float>(), y_te.data_ptr<float>(), n, key0,
                     key1, tau);
  int64_t n_test = total.to(at::kLong).item<int64_t>();
  int64_t n_train = n - n_test;
  return {X_tr.narrow(0, 0, n_train), y_tr.narrow(0, 0, n_train),
          X_te.narrow(0, 0, n_test), y_te.narrow(0, 0, n_test)};
}
