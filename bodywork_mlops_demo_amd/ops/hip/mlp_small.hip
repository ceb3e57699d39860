// MLP non-GEMM kernels: fused rank-1 expand, row-dot head, column reduce.
//
// MI355X-first design note (models/mlp.py docstring): the 1-feature input
// layer and the scalar output head are bandwidth-bound elementwise /
// reduction shapes — running them as K=1 / N=1 MFMA GEMMs would idle the
// matrix cores behind HBM, so they are dedicated fused kernels instead:
//   expand1d: out[i,j] = act(x[i]*w[j] + b[j]) [* (mask>0)]  (bf16 out)
//   rowdot:   out[i]   = sum_j h[i,j]*w[j] + b               (fp32 out)
//   coldot:   dw[j]    = sum_i m[i,j]*v[i] [, cs[j] = sum_i m[i,j]]
// All loads/stores are bf16x8 (16 B/lane) per Guideline 13.
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include "bf16_utils.h"
#include "reduce.h"

// ---- expand1d -------------------------------------------------------------

// HAS_MASK: multiply by a 1-BIT ReLU mask (uint8 [n, H/8], bit e of byte
// c = column 8c+e > 0) — 16x less mask traffic than re-reading the bf16
// activations, and the mask stays L2-resident at MLP scale.
// EMIT_MASK: write that bitmask for the activations this kernel produces.
template <bool RELU, bool HAS_BIAS, bool HAS_MASK, bool EMIT_MASK>
__global__ void expand1d_kernel(const float* __restrict__ x,
                                const bf16_t* __restrict__ w,
                                const bf16_t* __restrict__ b,
                                const unsigned char* __restrict__ mask,
                                unsigned char* __restrict__ mask_out,
                                bf16_t* __restrict__ out, long long n,
                                int h8 /* H/8 */) {
  const long long total = (long long)n * h8;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const long long row = idx / h8;
    const int c8 = (int)(idx % h8);
    float xv = x[row];
    float wv[8], bv[8], acc[8];
    bf16x8_to_f32(load_bf16x8(w + c8 * 8), wv);
    if (HAS_BIAS) bf16x8_to_f32(load_bf16x8(b + c8 * 8), bv);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      acc[e] = HAS_BIAS ? fmaf(xv, wv[e], bv[e]) : xv * wv[e];
      if (RELU) acc[e] = fmaxf(acc[e], 0.0f);
    }
    if (HAS_MASK) {
      unsigned char mb = mask[row * h8 + c8];
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] = (mb >> e) & 1 ? acc[e] : 0.0f;
    }
    if (EMIT_MASK) {
      unsigned char mb = 0;
#pragma unroll
      for (int e = 0; e < 8; ++e) mb |= (acc[e] > 0.0f ? 1u : 0u) << e;
      mask_out[row * h8 + c8] = mb;
    }
    store_bf16x8(out + row * (long long)h8 * 8 + c8 * 8, f32_to_bf16x8(acc));
  }
}

std::tuple<at::Tensor, at::Tensor> expand1d_bf16_hip(
    const at::Tensor& x, const at::Tensor& w,
    const c10::optional<at::Tensor>& b, bool relu,
    const c10::optional<at::Tensor>& mask, bool emit_mask) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kFloat);
  TORCH_CHECK(w.scalar_type() == at::kBFloat16);
  const long long n = x.numel();
  const long long H = w.numel();
  TORCH_CHECK(H % 8 == 0, "expand1d: H must be a multiple of 8");
  auto out = at::empty({n, H}, w.options());
  const bool has_bias = b.has_value();
  const bool has_mask = mask.has_value();
  at::Tensor mask_out;
  if (emit_mask)
    mask_out = at::empty({n, H / 8}, w.options().dtype(at::kByte));
  else
    mask_out = at::empty({0}, w.options().dtype(at::kByte));
  if (has_mask)
    TORCH_CHECK(mask->scalar_type() == at::kByte &&
                mask->numel() == n * (H / 8),
                "expand1d: mask must be uint8 [n, H/8] bitmask");
  auto stream = at::cuda::getCurrentCUDAStream();
  long long total = n * (H / 8);
  int grid = (int)std::min<long long>((total + 255) / 256, 2048);
  const bf16_t* wp = (const bf16_t*)w.data_ptr();
  const bf16_t* bp = has_bias ? (const bf16_t*)b->data_ptr() : nullptr;
  const unsigned char* mp =
      has_mask ? (const unsigned char*)mask->data_ptr() : nullptr;
  unsigned char* mop =
      emit_mask ? (unsigned char*)mask_out.data_ptr() : nullptr;

#define LAUNCH_E1D(R, B_, M_, E_)                                         \
  hipLaunchKernelGGL((expand1d_kernel<R, B_, M_, E_>), dim3(grid),        \
                     dim3(256), 0, stream, x.data_ptr<float>(), wp, bp,   \
                     mp, mop, (bf16_t*)out.data_ptr(), n, (int)(H / 8))
#define LAUNCH_E1D_E(R, B_, M_)                                           \
  do { if (emit_mask) LAUNCH_E1D(R, B_, M_, true);                        \
       else LAUNCH_E1D(R, B_, M_, false); } while (0)
  if (relu) {
    if (has_bias) { if (has_mask) LAUNCH_E1D_E(true, true, true); else LAUNCH_E1D_E(true, true, false); }
    else          { if (has_mask) LAUNCH_E1D_E(true, false, true); else LAUNCH_E1D_E(true, false, false); }
  } else {
    if (has_bias) { if (has_mask) LAUNCH_E1D_E(false, true, true); else LAUNCH_E1D_E(false, true, false); }
    else          { if (has_mask) LAUNCH_E1D_E(false, false, true); else LAUNCH_E1D_E(false, false, false); }
  }
#undef LAUNCH_E1D_E
#undef LAUNCH_E1D
  return {out, mask_out};
}

// ---- rowdot ---------------------------------------------------------------
// one wave per row; lane covers 8 columns per step (bf16x8)

__global__ void rowdot_kernel(const bf16_t* __restrict__ h,
                              const bf16_t* __restrict__ w,
                              float* __restrict__ out, long long n, int H,
                              const float* __restrict__ bias_p) {
  const float bias = *bias_p;  // device read: keeps predict capture-safe
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_per_block = blockDim.x >> 6;
  const long long row = (long long)blockIdx.x * waves_per_block + wave;
  if (row >= n) return;
  const bf16_t* hrow = h + row * (long long)H;
  float acc = 0.0f;
  for (int c = lane * 8; c < H; c += 64 * 8) {
    float hv[8], wv[8];
    bf16x8_to_f32(load_bf16x8(hrow + c), hv);
    bf16x8_to_f32(load_bf16x8(w + c), wv);
#pragma unroll
    for (int e = 0; e < 8; ++e) acc = fmaf(hv[e], wv[e], acc);
  }
  acc = wave_sum_f32(acc);
  if (lane == 0) out[row] = acc + bias;
}

at::Tensor rowdot_bf16_hip(const at::Tensor& h, const at::Tensor& w,
                           const at::Tensor& bias) {
  TORCH_CHECK(h.is_cuda() && h.dim() == 2);
  TORCH_CHECK(h.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16);
  TORCH_CHECK(bias.is_cuda() && bias.numel() == 1 &&
              bias.scalar_type() == at::kFloat);
  const long long n = h.size(0);
  const int H = (int)h.size(1);
  TORCH_CHECK(H % 8 == 0, "rowdot: H must be a multiple of 8");
  auto out = at::empty({n}, h.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const int waves_per_block = 4;
  long long grid = (n + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(rowdot_kernel, dim3(grid), dim3(64 * waves_per_block), 0,
                     stream, (const bf16_t*)h.data_ptr(),
                     (const bf16_t*)w.data_ptr(), out.data_ptr<float>(), n, H,
                     bias.data_ptr<float>());
  return out;
}

// ---- coldot (+ optional colsum) ------------------------------------------
// block covers 2048 consecutive columns (256 threads x 8); rows split
// across gridDim.y chunks; one fp32 atomicAdd per output per chunk.

template <bool ALSO_COLSUM>
__global__ void coldot_kernel(const bf16_t* __restrict__ m,
                              const float* __restrict__ v,
                              float* __restrict__ dw, float* __restrict__ cs,
                              long long n, int H) {
  const int col0 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (col0 >= H) return;
  const long long rows_per_chunk = (n + gridDim.y - 1) / gridDim.y;
  const long long r0 = blockIdx.y * rows_per_chunk;
  const long long r1 = min(r0 + rows_per_chunk, n);
  float dot[8] = {0}, sum[8] = {0};
  // 4-row unroll: independent 16-B loads in flight (single-row loop
  // measured latency-bound at ~2.8 TB/s)
  long long r = r0;
  for (; r + 3 < r1; r += 4) {
    bf16x8 raw[4];
    float vv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      raw[u] = load_bf16x8(m + (r + u) * (long long)H + col0);
      vv[u] = v[r + u];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      float mv[8];
      bf16x8_to_f32(raw[u], mv);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        dot[e] = fmaf(mv[e], vv[u], dot[e]);
        if (ALSO_COLSUM) sum[e] += mv[e];
      }
    }
  }
  for (; r < r1; ++r) {
    float vv = v[r];
    float mv[8];
    bf16x8_to_f32(load_bf16x8(m + r * (long long)H + col0), mv);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      dot[e] = fmaf(mv[e], vv, dot[e]);
      if (ALSO_COLSUM) sum[e] += mv[e];
    }
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    atomicAdd(&dw[col0 + e], dot[e]);
    if (ALSO_COLSUM) atomicAdd(&cs[col0 + e], sum[e]);
  }
}

at::Tensor coldot_bf16_hip(const at::Tensor& m, const at::Tensor& v,
                           bool also_colsum) {
  TORCH_CHECK(m.is_cuda() && m.dim() == 2);
  TORCH_CHECK(m.scalar_type() == at::kBFloat16 &&
              v.scalar_type() == at::kFloat);
  const long long n = m.size(0);
  const int H = (int)m.size(1);
  TORCH_CHECK(H % 8 == 0, "coldot: H must be a multiple of 8");
  auto out = at::zeros({also_colsum ? 2 : 1, H},
                       m.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  int gx = (H / 8 + 255) / 256;
  // >=4 waves/SIMD: 256 y-chunks gave only 1 workgroup/CU (4 waves) and
  // measured ~2-3.7 TB/s; 128-row chunks fill the chip (atomic fan-in
  // per column stays <=1024, negligible)
  int gy = (int)std::min<long long>((n + 127) / 128, 1024);
  float* dw = out.data_ptr<float>();
  float* cs = also_colsum ? dw + H : nullptr;
  if (also_colsum)
    hipLaunchKernelGGL((coldot_kernel<true>), dim3(gx, gy), dim3(256), 0,
                       stream, (const bf16_t*)m.data_ptr(),
                       v.data_ptr<float>(), dw, cs, n, H);
  else
    hipLaunchKernelGGL((coldot_kernel<false>), dim3(gx, gy), dim3(256), 0,
                       stream, (const bf16_t*)m.data_ptr(),
                       v.data_ptr<float>(), dw, nullptr, n, H);
  return out;
}

// colsum alone = coldot with v == 1 fused path
__global__ void colsum_kernel(const bf16_t* __restrict__ m,
                              float* __restrict__ cs, long long n, int H) {
  const int col0 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (col0 >= H) return;
  const long long rows_per_chunk = (n + gridDim.y - 1) / gridDim.y;
  const long long r0 = blockIdx.y * rows_per_chunk;
  const long long r1 = min(r0 + rows_per_chunk, n);
  float sum[8] = {0};
  long long r = r0;
  for (; r + 3 < r1; r += 4) {
    bf16x8 raw[4];
#pragma unroll
    for (int u = 0; u < 4; ++u)
      raw[u] = load_bf16x8(m + (r + u) * (long long)H + col0);
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      float mv[8];
      bf16x8_to_f32(raw[u], mv);
#pragma unroll
      for (int e = 0; e < 8; ++e) sum[e] += mv[e];
    }
  }
  for (; r < r1; ++r) {
    float mv[8];
    bf16x8_to_f32(load_bf16x8(m + r * (long long)H + col0), mv);
#pragma unroll
    for (int e = 0; e < 8; ++e) sum[e] += mv[e];
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) atomicAdd(&cs[col0 + e], sum[e]);
}

at::Tensor colsum_bf16_hip(const at::Tensor& m) {
  TORCH_CHECK(m.is_cuda() && m.dim() == 2 &&
              m.scalar_type() == at::kBFloat16);
  const long long n = m.size(0);
  const int H = (int)m.size(1);
  TORCH_CHECK(H % 8 == 0);
  auto out = at::zeros({H}, m.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  int gx = (H / 8 + 255) / 256;
  int gy = (int)std::min<long long>((n + 127) / 128, 1024);
  hipLaunchKernelGGL(colsum_kernel, dim3(gx, gy), dim3(256), 0, stream,
                     (const bf16_t*)m.data_ptr(), out.data_ptr<float>(), n, H);
  return out;
}
