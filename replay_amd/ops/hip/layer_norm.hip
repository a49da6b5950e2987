// Fused LayerNorm forward/backward for gfx950 (CDNA4).
//
// K3 in SURVEY §2.12: the reference runs torch LayerNorm(eps=1e-8) around
// every attention/FFN block (replay/nn/sequential/sasrec/transformer.py:47-61).
// Here: one 64-lane wave per row, fp32 accumulation, vectorized loads
// (guide G13: bf16 scalar loads are 2-2.5x slower than short4/short8).
// Rows are E = 64..1024 elements (d_model of recommender transformers), so a
// row fits a wave's registers; 4 waves per 256-thread workgroup process 4
// rows, grid-stride over N = B*L rows.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

template <typename T>
__global__ void ln_fwd_kernel(const T* __restrict__ x,
                              const float* __restrict__ weight,
                              const float* __restrict__ bias,
                              T* __restrict__ y,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out,
                              int64_t n_rows, int n_cols, float eps) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = (gridDim.x * blockDim.x) / WAVE;

  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const T* xr = x + row * n_cols;
    float sum = 0.f, sumsq = 0.f;
    for (int c = lane; c < n_cols; c += WAVE) {
      float v = to_f32<T>(xr[c]);
      sum += v;
      sumsq += v * v;
    }
    sum = wave_reduce_sum(sum);
    sumsq = wave_reduce_sum(sumsq);
    const float mean = sum / n_cols;
    const float var = sumsq / n_cols - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (lane == 0 && mean_out != nullptr) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    T* yr = y + row * n_cols;
    for (int c = lane; c < n_cols; c += WAVE) {
      float v = (to_f32<T>(xr[c]) - mean) * rstd;
      yr[c] = from_f32<T>(v * weight[c] + bias[c]);
    }
  }
}

// dx for one row needs two row-reductions of dy*w and dy*w*xhat.
// dweight/dbias are column reductions over all rows.  A per-element
// atomicAdd is catastrophic here (measured 1.25 ms per call at [51200,64]
// from contention on 64 fp32 words): instead each wave accumulates its
// grid-stride rows' partials in registers (E <= 8*WAVE columns per lane) and
// issues ONE atomicAdd per column at the end (guide Guideline 12).
#define LN_MAX_COLS_PER_LANE 8  // supports E up to 512

template <typename T>
__global__ void ln_bwd_kernel(const T* __restrict__ x,
                              const T* __restrict__ dy,
                              const float* __restrict__ weight,
                              const float* __restrict__ mean,
                              const float* __restrict__ rstd,
                              T* __restrict__ dx,
                              float* __restrict__ dweight,
                              float* __restrict__ dbias,
                              int64_t n_rows, int n_cols) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = (gridDim.x * blockDim.x) / WAVE;

  float acc_dw[LN_MAX_COLS_PER_LANE];
  float acc_db[LN_MAX_COLS_PER_LANE];
#pragma unroll
  for (int i = 0; i < LN_MAX_COLS_PER_LANE; ++i) acc_dw[i] = acc_db[i] = 0.f;

  // two rows per iteration: independent latency chains (the single-row form
  // measured latency-bound at 90 us for [51200, 64] fp32, ~0.4 TB/s)
  const int64_t pair_stride = (int64_t)n_waves * 2;
  for (int64_t row = (int64_t)wave_id * 2; row < n_rows; row += pair_stride) {
    const bool has2 = row + 1 < n_rows;
    const T* xr0 = x + row * n_cols;
    const T* dyr0 = dy + row * n_cols;
    const T* xr1 = xr0 + (has2 ? n_cols : 0);
    const T* dyr1 = dyr0 + (has2 ? n_cols : 0);
    const float m0 = mean[row], rs0 = rstd[row];
    const float m1 = mean[row + (has2 ? 1 : 0)], rs1 = rstd[row + (has2 ? 1 : 0)];
    float s1a = 0.f, s2a = 0.f, s1b = 0.f, s2b = 0.f;
    for (int c = lane; c < n_cols; c += WAVE) {
      const float wgt = weight[c];
      float xh0 = (to_f32<T>(xr0[c]) - m0) * rs0;
      float dw0 = to_f32<T>(dyr0[c]) * wgt;
      float xh1 = (to_f32<T>(xr1[c]) - m1) * rs1;
      float dw1 = to_f32<T>(dyr1[c]) * wgt;
      s1a += dw0; s2a += dw0 * xh0;
      s1b += dw1; s2b += dw1 * xh1;
    }
    s1a = wave_reduce_sum(s1a) / n_cols;
    s2a = wave_reduce_sum(s2a) / n_cols;
    s1b = wave_reduce_sum(s1b) / n_cols;
    s2b = wave_reduce_sum(s2b) / n_cols;
    T* dxr0 = dx + row * n_cols;
    T* dxr1 = dxr0 + n_cols;
    int i = 0;
    for (int c = lane; c < n_cols; c += WAVE, ++i) {
      const float wgt = weight[c];
      float xh0 = (to_f32<T>(xr0[c]) - m0) * rs0;
      float dy0 = to_f32<T>(dyr0[c]);
      dxr0[c] = from_f32<T>((dy0 * wgt - s1a - xh0 * s2a) * rs0);
      acc_dw[i] += dy0 * xh0;
      acc_db[i] += dy0;
      if (has2) {
        float xh1 = (to_f32<T>(xr1[c]) - m1) * rs1;
        float dy1 = to_f32<T>(dyr1[c]);
        dxr1[c] = from_f32<T>((dy1 * wgt - s1b - xh1 * s2b) * rs1);
        acc_dw[i] += dy1 * xh1;
        acc_db[i] += dy1;
      }
    }
  }
  int i = 0;
  for (int c = lane; c < n_cols; c += WAVE, ++i) {
    atomicAdd(&dweight[c], acc_dw[i]);
    atomicAdd(&dbias[c], acc_db[i]);
  }
}

template <typename T>
void ln_fwd_launch(const torch::Tensor& x, const torch::Tensor& w, const torch::Tensor& b,
                   torch::Tensor& y, torch::Tensor& mean, torch::Tensor& rstd, double eps) {
  const int64_t n_rows = x.numel() / x.size(-1);
  const int n_cols = x.size(-1);
  const int threads = 256;
  const int waves_per_block = threads / WAVE;
  // >> 256 workgroups to fill 256 CUs / 8 XCDs (guide §1)
  int blocks = (int)std::min<int64_t>((n_rows + waves_per_block - 1) / waves_per_block, 8192);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(ln_fwd_kernel<T>, dim3(blocks), dim3(threads), 0, stream,
                     reinterpret_cast<const T*>(x.data_ptr()), w.data_ptr<float>(),
                     b.data_ptr<float>(), reinterpret_cast<T*>(y.data_ptr()),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), n_rows, n_cols,
                     (float)eps);
}

template <typename T>
void ln_bwd_launch(const torch::Tensor& x, const torch::Tensor& dy, const torch::Tensor& w,
                   const torch::Tensor& mean, const torch::Tensor& rstd, torch::Tensor& dx,
                   torch::Tensor& dw, torch::Tensor& db) {
  const int64_t n_rows = x.numel() / x.size(-1);
  const int n_cols = x.size(-1);
  TORCH_CHECK(n_cols <= WAVE * LN_MAX_COLS_PER_LANE, "LN bwd supports E<=512");
  const int threads = 256;
  const int waves_per_block = threads / WAVE;
  // cap waves: each wave does one atomicAdd per column at the end, so more
  // waves = more atomic traffic; 1024 blocks = 4096 waves (rows split in pairs)
  int blocks = (int)std::min<int64_t>((n_rows / 2 + waves_per_block - 1) / waves_per_block, 1024);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(ln_bwd_kernel<T>, dim3(blocks), dim3(threads), 0, stream,
                     reinterpret_cast<const T*>(x.data_ptr()),
                     reinterpret_cast<const T*>(dy.data_ptr()), w.data_ptr<float>(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     reinterpret_cast<T*>(dx.data_ptr()), dw.data_ptr<float>(),
                     db.data_ptr<float>(), n_rows, n_cols);
}

}  // namespace

std::vector<torch::Tensor> layer_norm_fwd(torch::Tensor x, torch::Tensor weight,
                                          torch::Tensor bias, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be contiguous CUDA");
  auto w = weight.to(torch::kFloat32).contiguous();
  auto b = bias.to(torch::kFloat32).contiguous();
  auto y = torch::empty_like(x);
  const int64_t n_rows = x.numel() / x.size(-1);
  auto opts = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({n_rows}, opts);
  auto rstd = torch::empty({n_rows}, opts);
  if (x.scalar_type() == torch::kBFloat16) {
    ln_fwd_launch<__hip_bfloat16>(x, w, b, y, mean, rstd, eps);
  } else if (x.scalar_type() == torch::kHalf) {
    ln_fwd_launch<__half>(x, w, b, y, mean, rstd, eps);
  } else if (x.scalar_type() == torch::kFloat32) {
    ln_fwd_launch<float>(x, w, b, y, mean, rstd, eps);
  } else {
    TORCH_CHECK(false, "unsupported dtype");
  }
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layer_norm_bwd(torch::Tensor x, torch::Tensor dy,
                                          torch::Tensor weight, torch::Tensor mean,
                                          torch::Tensor rstd) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  auto w = weight.to(torch::kFloat32).contiguous();
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({x.size(-1)}, x.options().dtype(torch::kFloat32));
  auto db = torch::zeros({x.size(-1)}, x.options().dtype(torch::kFloat32));
  if (x.scalar_type() == torch::kBFloat16) {
    ln_bwd_launch<__hip_bfloat16>(x, dy, w, mean, rstd, dx, dw, db);
  } else if (x.scalar_type() == torch::kHalf) {
    ln_bwd_launch<__half>(x, dy, w, mean, rstd, dx, dw, db);
  } else if (x.scalar_type() == torch::kFloat32) {
    ln_bwd_launch<float>(x, dy, w, mean, rstd, dx, dw, db);
  } else {
    TORCH_CHECK(false, "unsupported dtype");
  }
  return {dx, dw, db};
}
