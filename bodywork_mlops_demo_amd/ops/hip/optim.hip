// Fused optimizer + weight-copy kernels for the MLP training loop.
//
//   adam_step:        in-place fp32 Adam update (bias-corrected), fused
//                     with the bf16 shadow-weight write the next forward
//                     needs — one pass over the parameter instead of the
//                     ~6 eager elementwise launches of a torch Adam step.
//   transpose_to_bf16: 32x32 LDS-tiled fp32->bf16 transpose producing the
//                     [out,in] / [in,out] weight pair the NT MFMA GEMM
//                     consumes (both copies K-contiguous, gemm.hip).
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include "bf16_utils.h"
#include "philox.h"

template <bool WRITE_BF16>
__global__ void adam_step_kernel(float* __restrict__ p, const float* __restrict__ g,
                                 float* __restrict__ m, float* __restrict__ v,
                                 bf16_t* __restrict__ p_bf16, long long n,
                                 float lr, float beta1, float beta2, float eps,
                                 float inv_bc1, float inv_bc2,
                                 const float* __restrict__ bc /* nullable */) {
  if (bc) {  // device-side bias correction: hipGraph-replayable steps
    inv_bc1 = bc[0];
    inv_bc2 = bc[1];
  }
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float gv = g[i];
    float mv = beta1 * m[i] + (1.0f - beta1) * gv;
    float vv = beta2 * v[i] + (1.0f - beta2) * gv * gv;
    m[i] = mv;
    v[i] = vv;
    float mhat = mv * inv_bc1;
    float vhat = vv * inv_bc2;
    float pv = p[i] - lr * mhat / (sqrtf(vhat) + eps);
    p[i] = pv;
    if (WRITE_BF16) p_bf16[i] = f32_to_bf16(pv);
  }
}

// device-side minibatch sampling: philox-keyed indices + a step counter
// that advances on device, so a captured training step resamples on every
// graph replay (no host RNG, no H2D index copy)
__global__ void batch_indices_kernel(int64_t* __restrict__ out,
                                     unsigned long long* __restrict__ ctr,
                                     const int64_t* __restrict__ n_ptr,
                                     long long bs, unsigned int key0,
                                     unsigned int key1) {
  const unsigned long long step = *ctr;
  const unsigned long long n_data = (unsigned long long)*n_ptr;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < bs; i += stride) {
    Philox4 r = philox4x32(step * (unsigned long long)bs + i, key0, key1);
    out[i] = (int64_t)(((unsigned long long)r.x << 32 | r.y) % n_data);
  }
}

__global__ void bump_counter_kernel(unsigned long long* ctr) { ++*ctr; }

at::Tensor batch_indices_hip(at::Tensor ctr, const at::Tensor& n_dev,
                             int64_t bs, int64_t seed) {
  TORCH_CHECK(ctr.is_cuda() && ctr.numel() == 1 &&
              ctr.scalar_type() == at::kLong);
  TORCH_CHECK(n_dev.is_cuda() && n_dev.numel() == 1 &&
              n_dev.scalar_type() == at::kLong);
  auto out = at::empty({bs}, ctr.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  unsigned int key0 = (unsigned int)(seed & 0xFFFFFFFFll);
  int grid = (int)std::min<int64_t>((bs + 255) / 256, 1024);
  hipLaunchKernelGGL(batch_indices_kernel, dim3(grid), dim3(256), 0, stream,
                     out.data_ptr<int64_t>(),
                     (unsigned long long*)ctr.data_ptr(),
                     n_dev.data_ptr<int64_t>(), bs, key0, 0xB5297A4Du);
  hipLaunchKernelGGL(bump_counter_kernel, dim3(1), dim3(1), 0, stream,
                     (unsigned long long*)ctr.data_ptr());
  return out;
}

void adam_step_hip(at::Tensor p, const at::Tensor& g, at::Tensor m,
                   at::Tensor v, const c10::optional<at::Tensor>& p_bf16,
                   double lr, double beta1, double beta2, double eps,
                   int64_t t, const c10::optional<at::Tensor>& bc) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == at::kFloat);
  TORCH_CHECK(g.numel() == p.numel() && m.numel() == p.numel() &&
              v.numel() == p.numel());
  long long n = p.numel();
  float inv_bc1 = 1.0f / (1.0f - powf((float)beta1, (float)t));
  float inv_bc2 = 1.0f / (1.0f - powf((float)beta2, (float)t));
  const float* bcp = nullptr;
  if (bc.has_value()) {
    TORCH_CHECK(bc->numel() == 2 && bc->scalar_type() == at::kFloat &&
                bc->is_cuda());
    bcp = bc->data_ptr<float>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = (int)std::min<long long>((n + 1023) / 1024, 2048);
  if (p_bf16.has_value()) {
    TORCH_CHECK(p_bf16->numel() == n &&
                p_bf16->scalar_type() == at::kBFloat16);
    hipLaunchKernelGGL((adam_step_kernel<true>), dim3(grid), dim3(256), 0,
                       stream, p.data_ptr<float>(), g.data_ptr<float>(),
                       m.data_ptr<float>(), v.data_ptr<float>(),
                       (bf16_t*)p_bf16->data_ptr(), n, (float)lr,
                       (float)beta1, (float)beta2, (float)eps, inv_bc1,
                       inv_bc2, bcp);
  } else {
    hipLaunchKernelGGL((adam_step_kernel<false>), dim3(grid), dim3(256), 0,
                       stream, p.data_ptr<float>(), g.data_ptr<float>(),
                       m.data_ptr<float>(), v.data_ptr<float>(), nullptr, n,
                       (float)lr, (float)beta1, (float)beta2, (float)eps,
                       inv_bc1, inv_bc2, bcp);
  }
}

// 32x32 LDS-tiled transpose, fp32 source -> bf16 transposed destination.
// +1 padding column breaks the write-side bank conflict (Guideline 4).
__global__ void transpose_to_bf16_kernel(const float* __restrict__ src,
                                         bf16_t* __restrict__ dst,
                                         int rows, int cols) {
  __shared__ float tile[32][33];
  int c0 = blockIdx.x * 32;
  int r0 = blockIdx.y * 32;
  // 256 threads: 8 rows of 32 per pass, 4 passes
  int tc = threadIdx.x & 31;
  int tr = threadIdx.x >> 5;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    int r = r0 + tr + p * 8;
    int c = c0 + tc;
    tile[tr + p * 8][tc] = (r < rows && c < cols) ? src[(long long)r * cols + c]
                                                  : 0.0f;
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    int r = c0 + tr + p * 8;  // destination row = source col
    int c = r0 + tc;          // destination col = source row
    if (r < cols && c < rows)
      dst[(long long)r * rows + c] = f32_to_bf16(tile[tc][tr + p * 8]);
  }
}

// 64x64 LDS-tiled bf16->bf16 transpose, 16-B loads and stores.
// Feeds the dW = dY^T @ X shape to the fast NT GEMM: two bandwidth-bound
// transposes (~0.5 ms for a 65536x4096 operand pair) instead of the
// scalar transpose-staged TN kernel (7x slower at H=4096).
__global__ void transpose_bf16_kernel(const bf16_t* __restrict__ src,
                                      bf16_t* __restrict__ dst,
                                      long long rows, long long cols) {
  __shared__ short tile[64][72];  // +8 pad: conflict-free column reads
  const long long r0 = (long long)blockIdx.y * 64;
  const long long c0 = (long long)blockIdx.x * 64;
  const int tc = threadIdx.x & 7;   // 16-B chunk within the row
  const int tr = threadIdx.x >> 3;  // row within the tile (32 per pass)
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    long long r = r0 + tr + p * 32;
    long long c = c0 + tc * 8;
    lds_vec v;
    if (r < rows && c + 7 < cols) {
      v = *(const lds_vec*)(src + r * cols + c);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e)
        v.s[e] = (r < rows && c + e < cols)
                     ? ((const short*)src)[r * cols + c + e]
                     : (short)0;
    }
    *(lds_vec*)&tile[tr + p * 32][tc * 8] = v;
  }
  __syncthreads();
  // write transposed: thread handles out row (= src col) c0+tr(+32p),
  // 8 consecutive out cols (= src rows) r0+tc*8..
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    long long oc = c0 + tr + p * 32;   // out row = src col
    long long orow0 = r0 + tc * 8;     // out col base = src row
    if (oc >= cols) continue;
    lds_vec v;
#pragma unroll
    for (int e = 0; e < 8; ++e) v.s[e] = tile[tc * 8 + e][tr + p * 32];
    if (orow0 + 7 < rows) {
      *(lds_vec*)(dst + oc * rows + orow0) = v;
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e)
        if (orow0 + e < rows)
          ((short*)dst)[oc * rows + orow0 + e] = v.s[e];
    }
  }
}

at::Tensor transpose_bf16_hip(const at::Tensor& src) {
  TORCH_CHECK(src.is_cuda() && src.dim() == 2 &&
              src.scalar_type() == at::kBFloat16);
  long long rows = src.size(0), cols = src.size(1);
  auto dst = at::empty({cols, rows}, src.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  dim3 grid((cols + 63) / 64, (rows + 63) / 64);
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, stream,
                     (const bf16_t*)src.data_ptr(), (bf16_t*)dst.data_ptr(),
                     rows, cols);
  return dst;
}

at::Tensor transpose_to_bf16_hip(const at::Tensor& src) {
  TORCH_CHECK(src.is_cuda() && src.dim() == 2 &&
              src.scalar_type() == at::kFloat);
  int rows = (int)src.size(0), cols = (int)src.size(1);
  auto dst = at::empty({cols, rows}, src.options().dtype(at::kBFloat16));
  auto stream = at::cuda::getCurrentCUDAStream();
  dim3 grid((cols + 31) / 32, (rows + 31) / 32);
  hipLaunchKernelGGL(transpose_to_bf16_kernel, grid, dim3(256), 0, stream,
                     src.data_ptr<float>(), (bf16_t*)dst.data_ptr(), rows,
                     cols);
  return dst;
}
