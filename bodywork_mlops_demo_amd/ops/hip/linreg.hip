// Fused OLS statistics, batched scoring and metric reductions — gfx950.
//
// Replaces sklearn LinearRegression.fit / predict and the sklearn metric
// calls (reference stage_1:79-108, stage_2:78, stage_4:101-113) with:
//   - linreg_stats:   single pass over (X, y) producing fp64
//                     [n, sum_x, sum_y, sum_xx, sum_xy] — the whole
//                     closed-form fit input AND the whole DP all-reduce
//                     payload.
//   - linear_score:   yhat = intercept + coef*X, float4-vectorised,
//                     grid-stride (hipGraph-capturable: no allocs/syncs).
//   - regression_metrics / score_label_metrics: one fused pass producing
//     every sum the MAPE/R^2/max-residual/Pearson formulas need.
// All reductions: per-wave shuffle -> LDS -> one fp64 atomicAdd per block
// (Guideline 12); accumulation in fp64 so 1B-row sums stay exact to ~2^-40.
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include "reduce.h"

#define RED_BLOCK 256
#define RED_WAVES (RED_BLOCK / 64)

static inline int reduce_grid(long long n, int per_thread = 4) {
  long long want = (n + (long long)RED_BLOCK * per_thread - 1) /
                   ((long long)RED_BLOCK * per_thread);
  return (int)std::min<long long>(want, 2048);  // Guideline 11 cap
}

// ---- linreg_stats ---------------------------------------------------------

__global__ void linreg_stats_kernel(const float* __restrict__ x,
                                    const float* __restrict__ y,
                                    double* __restrict__ out, long long n) {
  // 4 independent partial accumulators per statistic (one per float4
  // lane) break the per-iteration fp64 dependency chains that left the
  // serial version at ~12% of HBM bandwidth
  double sx[4] = {0, 0, 0, 0}, sy[4] = {0, 0, 0, 0};
  double sxx[4] = {0, 0, 0, 0}, sxy[4] = {0, 0, 0, 0};
  const long long stride = (long long)gridDim.x * RED_BLOCK;
  long long i = (long long)blockIdx.x * RED_BLOCK + threadIdx.x;
  const long long n4 = n & ~3ll;
  for (long long j = i * 4; j < n4; j += stride * 4) {
    float4 xv = *(const float4*)(x + j);
    float4 yv = *(const float4*)(y + j);
    const float xs[4] = {xv.x, xv.y, xv.z, xv.w};
    const float ys[4] = {yv.x, yv.y, yv.z, yv.w};
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      double xd = xs[e], yd = ys[e];
      sx[e] += xd;
      sy[e] += yd;
      sxx[e] = fma(xd, xd, sxx[e]);
      sxy[e] = fma(xd, yd, sxy[e]);
    }
  }
  for (long long j = n4 + i; j < n; j += stride) {
    double xv = x[j], yv = y[j];
    sx[0] += xv; sy[0] += yv;
    sxx[0] = fma(xv, xv, sxx[0]);
    sxy[0] = fma(xv, yv, sxy[0]);
  }
#pragma unroll
  for (int e = 1; e < 4; ++e) {
    sx[0] += sx[e]; sy[0] += sy[e]; sxx[0] += sxx[e]; sxy[0] += sxy[e];
  }
  __shared__ double lds[RED_WAVES];
  double t;
  t = block_sum_f64<RED_WAVES>(sx[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[1], t);
  t = block_sum_f64<RED_WAVES>(sy[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[2], t);
  t = block_sum_f64<RED_WAVES>(sxx[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[3], t);
  t = block_sum_f64<RED_WAVES>(sxy[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[4], t);
}

at::Tensor linreg_stats_hip(const at::Tensor& x, const at::Tensor& y) {
  TORCH_CHECK(x.is_cuda() && y.is_cuda() && x.numel() == y.numel());
  TORCH_CHECK(x.scalar_type() == at::kFloat && y.scalar_type() == at::kFloat);
  long long n = x.numel();
  auto out = at::zeros({5}, x.options().dtype(at::kDouble));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(linreg_stats_kernel, dim3(reduce_grid(n)),
                     dim3(RED_BLOCK), 0, stream, x.data_ptr<float>(),
                     y.data_ptr<float>(), out.data_ptr<double>(), n);
  out[0].fill_((double)n);
  return out;
}

// ---- linear_score ---------------------------------------------------------

__global__ void linear_score_kernel(const float* __restrict__ x,
                                    float* __restrict__ out,
                                    const float* __restrict__ ab,
                                    long long n) {
  // coefficients read from device memory so a captured hipGraph picks up
  // redeployed models without recapture (serving hot-swap)
  const float a = ab[0], b = ab[1];
  const long long stride = (long long)gridDim.x * blockDim.x;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long n4 = n >> 2;
  const float4* x4 = (const float4*)x;
  float4* o4 = (float4*)out;
  for (long long j = i; j < n4; j += stride) {
    float4 v = x4[j];
    o4[j] = {fmaf(b, v.x, a), fmaf(b, v.y, a), fmaf(b, v.z, a),
             fmaf(b, v.w, a)};
  }
  for (long long j = n4 * 4 + i; j < n; j += stride)
    out[j] = fmaf(b, x[j], a);
}

at::Tensor linear_score_hip(const at::Tensor& x, const at::Tensor& ab) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kFloat);
  TORCH_CHECK(ab.is_cuda() && ab.numel() == 2 &&
              ab.scalar_type() == at::kFloat);
  long long n = x.numel();
  auto out = at::empty_like(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(linear_score_kernel, dim3(reduce_grid(n)),
                     dim3(RED_BLOCK), 0, stream, x.data_ptr<float>(),
                     out.data_ptr<float>(), ab.data_ptr<float>(), n);
  return out;
}

// ---- polynomial ridge: fused X^T X / X^T y over an IMPLICIT basis --------
// The k-feature generalisation of linreg_stats promised by the SURVEY
// mapping table: sufficient statistics for ridge/OLS over the normalised
// polynomial basis phi_j(x) = t^j, t = (x - mu)/s, j = 0..NF-1.  The
// design matrix is never materialised — each thread expands its rows'
// basis on the fly and accumulates the upper-triangular X^T X plus X^T y
// in fp64, so the pass reads exactly the 8 bytes/row the linear fit
// reads regardless of degree.  Output layout:
//   out[0]                    = n
//   out[1 .. T]               = upper triangle of X^T X, row-major
//                               (T = NF*(NF+1)/2)
//   out[T+1 .. T+NF]          = X^T y
// Host side solves the (ridge-regularised) normal equations.

template <int NF>
__global__ void poly_stats_kernel(const float* __restrict__ x,
                                  const float* __restrict__ y,
                                  double* __restrict__ out, long long n,
                                  float mu, float inv_s) {
  constexpr int TRI = NF * (NF + 1) / 2;
  double acc[TRI + NF];
#pragma unroll
  for (int i = 0; i < TRI + NF; ++i) acc[i] = 0.0;
  const long long stride = (long long)gridDim.x * RED_BLOCK;
  for (long long j = (long long)blockIdx.x * RED_BLOCK + threadIdx.x; j < n;
       j += stride) {
    double t = (double)((x[j] - mu) * inv_s);
    double yv = y[j];
    double phi[NF];
    phi[0] = 1.0;
#pragma unroll
    for (int p = 1; p < NF; ++p) phi[p] = phi[p - 1] * t;
    int s = 0;
#pragma unroll
    for (int a = 0; a < NF; ++a) {
#pragma unroll
      for (int b = a; b < NF; ++b) acc[s++] += phi[a] * phi[b];
    }
#pragma unroll
    for (int a = 0; a < NF; ++a) acc[TRI + a] += phi[a] * yv;
  }
  __shared__ double lds[RED_WAVES];
#pragma unroll
  for (int i = 0; i < TRI + NF; ++i) {
    double v = block_sum_f64<RED_WAVES>(acc[i], lds);
    if (threadIdx.x == 0) atomicAdd(&out[1 + i], v);
  }
}

at::Tensor poly_stats_hip(const at::Tensor& x, const at::Tensor& y,
                          int64_t nf, double mu, double s) {
  TORCH_CHECK(x.is_cuda() && y.is_cuda() && x.numel() == y.numel());
  TORCH_CHECK(x.scalar_type() == at::kFloat && y.scalar_type() == at::kFloat);
  TORCH_CHECK(nf >= 2 && nf <= 6, "poly_stats: 2 <= degree+1 <= 6");
  long long n = x.numel();
  int tri = (int)(nf * (nf + 1) / 2);
  auto out = at::zeros({1 + tri + nf}, x.options().dtype(at::kDouble));
  auto stream = at::cuda::getCurrentCUDAStream();
  dim3 grid(reduce_grid(n));
  float inv_s = (float)(1.0 / s);
#define PS_LAUNCH(NF_)                                                     \
  hipLaunchKernelGGL((poly_stats_kernel<NF_>), grid, dim3(RED_BLOCK), 0,   \
                     stream, x.data_ptr<float>(), y.data_ptr<float>(),     \
                     out.data_ptr<double>(), n, (float)mu, inv_s)
  switch (nf) {
    case 2: PS_LAUNCH(2); break;
    case 3: PS_LAUNCH(3); break;
    case 4: PS_LAUNCH(4); break;
    case 5: PS_LAUNCH(5); break;
    default: PS_LAUNCH(6); break;
  }
#undef PS_LAUNCH
  out[0].fill_((double)n);
  return out;
}

// Horner-scheme polynomial scoring in the normalised basis; coefficients
// read from device memory so captured serving graphs follow redeploys.
__global__ void poly_score_kernel(const float* __restrict__ x,
                                  float* __restrict__ outv,
                                  const float* __restrict__ coef, int nf,
                                  float mu, float inv_s, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long j = (long long)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += stride) {
    float t = (x[j] - mu) * inv_s;
    float acc = coef[nf - 1];
    for (int p = nf - 2; p >= 0; --p) acc = fmaf(acc, t, coef[p]);
    outv[j] = acc;
  }
}

at::Tensor poly_score_hip(const at::Tensor& x, const at::Tensor& coef,
                          double mu, double s) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kFloat);
  TORCH_CHECK(coef.is_cuda() && coef.scalar_type() == at::kFloat &&
              coef.numel() >= 2);
  long long n = x.numel();
  auto out = at::empty_like(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(poly_score_kernel, dim3(reduce_grid(n)),
                     dim3(RED_BLOCK), 0, stream, x.data_ptr<float>(),
                     out.data_ptr<float>(), coef.data_ptr<float>(),
                     (int)coef.numel(), (float)mu, (float)(1.0 / s), n);
  return out;
}

// ---- regression_metrics (offline: stage_1:79-90 semantics) ---------------
// out = [n, sum_ape, ss_res, sum_y, sum_yy, max_resid]

__global__ void regression_metrics_kernel(const float* __restrict__ y,
                                          const float* __restrict__ yhat,
                                          double* __restrict__ out,
                                          long long n) {
  const double MAPE_EPS = 2.220446049250313e-16;  // sklearn epsilon
  double s_ape[4] = {0, 0, 0, 0}, ss_res[4] = {0, 0, 0, 0};
  double s_y[4] = {0, 0, 0, 0}, s_yy[4] = {0, 0, 0, 0};
  double max_res = 0;
  const long long stride = (long long)gridDim.x * RED_BLOCK;
  const long long i = (long long)blockIdx.x * RED_BLOCK + threadIdx.x;
  const long long n4 = n & ~3ll;
  for (long long j = i * 4; j < n4; j += stride * 4) {
    float4 yv4 = *(const float4*)(y + j);
    float4 pv4 = *(const float4*)(yhat + j);
    const float ys[4] = {yv4.x, yv4.y, yv4.z, yv4.w};
    const float ps[4] = {pv4.x, pv4.y, pv4.z, pv4.w};
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      double yv = ys[e], r = (double)ys[e] - ps[e];
      double ar = fabs(r);
      s_ape[e] += ar / fmax(fabs(yv), MAPE_EPS);
      ss_res[e] = fma(r, r, ss_res[e]);
      s_y[e] += yv;
      s_yy[e] = fma(yv, yv, s_yy[e]);
      max_res = fmax(max_res, ar);
    }
  }
  for (long long j = n4 + i; j < n; j += stride) {
    double yv = y[j], r = yv - yhat[j];
    double ar = fabs(r);
    s_ape[0] += ar / fmax(fabs(yv), MAPE_EPS);
    ss_res[0] = fma(r, r, ss_res[0]);
    s_y[0] += yv;
    s_yy[0] = fma(yv, yv, s_yy[0]);
    max_res = fmax(max_res, ar);
  }
#pragma unroll
  for (int e = 1; e < 4; ++e) {
    s_ape[0] += s_ape[e]; ss_res[0] += ss_res[e];
    s_y[0] += s_y[e]; s_yy[0] += s_yy[e];
  }
  __shared__ double lds[RED_WAVES];
  double t;
  t = block_sum_f64<RED_WAVES>(s_ape[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[1], t);
  t = block_sum_f64<RED_WAVES>(ss_res[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[2], t);
  t = block_sum_f64<RED_WAVES>(s_y[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[3], t);
  t = block_sum_f64<RED_WAVES>(s_yy[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[4], t);
  max_res = wave_max_f64(max_res);
  if ((threadIdx.x & 63) == 0) atomic_max_nonneg_f64(&out[5], max_res);
}

at::Tensor regression_metrics_hip(const at::Tensor& y,
                                  const at::Tensor& yhat) {
  TORCH_CHECK(y.is_cuda() && yhat.is_cuda() && y.numel() == yhat.numel());
  long long n = y.numel();
  auto yf = y.scalar_type() == at::kFloat ? y : y.to(at::kFloat);
  auto pf = yhat.scalar_type() == at::kFloat ? yhat : yhat.to(at::kFloat);
  auto out = at::zeros({6}, y.options().dtype(at::kDouble));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(regression_metrics_kernel, dim3(reduce_grid(n)),
                     dim3(RED_BLOCK), 0, stream, yf.data_ptr<float>(),
                     pf.data_ptr<float>(), out.data_ptr<double>(), n);
  out[0].fill_((double)n);
  return out;
}

// ---- score_label_metrics (online: stage_4:87-113 semantics) ---------------
// out = [n, sum_ape, max_ape, s_s, s_l, s_ss, s_ll, s_sl]

__global__ void score_label_metrics_kernel(const float* __restrict__ s,
                                           const float* __restrict__ l,
                                           double* __restrict__ out,
                                           long long n) {
  double s_ape[4] = {0, 0, 0, 0}, s_s[4] = {0, 0, 0, 0};
  double s_l[4] = {0, 0, 0, 0}, s_ss[4] = {0, 0, 0, 0};
  double s_ll[4] = {0, 0, 0, 0}, s_sl[4] = {0, 0, 0, 0};
  double max_ape = 0;
  const long long stride = (long long)gridDim.x * RED_BLOCK;
  const long long i = (long long)blockIdx.x * RED_BLOCK + threadIdx.x;
  const long long n4 = n & ~3ll;
  for (long long j = i * 4; j < n4; j += stride * 4) {
    float4 sv4 = *(const float4*)(s + j);
    float4 lv4 = *(const float4*)(l + j);
    const float ss4[4] = {sv4.x, sv4.y, sv4.z, sv4.w};
    const float ll4[4] = {lv4.x, lv4.y, lv4.z, lv4.w};
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      double sv = ss4[e], lv = ll4[e];
      double ape = fabs(sv / lv - 1.0);  // reference has no eps guard
      s_ape[e] += ape;
      max_ape = fmax(max_ape, ape);
      s_s[e] += sv;
      s_l[e] += lv;
      s_ss[e] = fma(sv, sv, s_ss[e]);
      s_ll[e] = fma(lv, lv, s_ll[e]);
      s_sl[e] = fma(sv, lv, s_sl[e]);
    }
  }
  for (long long j = n4 + i; j < n; j += stride) {
    double sv = s[j], lv = l[j];
    double ape = fabs(sv / lv - 1.0);
    s_ape[0] += ape;
    max_ape = fmax(max_ape, ape);
    s_s[0] += sv; s_l[0] += lv;
    s_ss[0] = fma(sv, sv, s_ss[0]);
    s_ll[0] = fma(lv, lv, s_ll[0]);
    s_sl[0] = fma(sv, lv, s_sl[0]);
  }
#pragma unroll
  for (int e = 1; e < 4; ++e) {
    s_ape[0] += s_ape[e]; s_s[0] += s_s[e]; s_l[0] += s_l[e];
    s_ss[0] += s_ss[e]; s_ll[0] += s_ll[e]; s_sl[0] += s_sl[e];
  }
  __shared__ double lds[RED_WAVES];
  double t;
  t = block_sum_f64<RED_WAVES>(s_ape[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[1], t);
  t = block_sum_f64<RED_WAVES>(s_s[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[3], t);
  t = block_sum_f64<RED_WAVES>(s_l[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[4], t);
  t = block_sum_f64<RED_WAVES>(s_ss[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[5], t);
  t = block_sum_f64<RED_WAVES>(s_ll[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[6], t);
  t = block_sum_f64<RED_WAVES>(s_sl[0], lds);
  if (threadIdx.x == 0) atomicAdd(&out[7], t);
  max_ape = wave_max_f64(max_ape);
  if ((threadIdx.x & 63) == 0) atomic_max_nonneg_f64(&out[2], max_ape);
}

at::Tensor score_label_metrics_hip(const at::Tensor& s, const at::Tensor& l) {
  TORCH_CHECK(s.is_cuda() && l.is_cuda() && s.numel() == l.numel());
  long long n = s.numel();
  auto sf = s.scalar_type() == at::kFloat ? s : s.to(at::kFloat);
  auto lf = l.scalar_type() == at::kFloat ? l : l.to(at::kFloat);
  auto out = at::zeros({8}, s.options().dtype(at::kDouble));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(score_label_metrics_kernel, dim3(reduce_grid(n)),
                     dim3(RED_BLOCK), 0, stream, sf.data_ptr<float>(),
                     lf.data_ptr<float>(), out.data_ptr<double>(), n);
  out[0].fill_((double)n);
  return out;
}
