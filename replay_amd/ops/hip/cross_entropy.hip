// Fused softmax cross-entropy over large catalogs for gfx950 (CDNA4).
//
// K10 in SURVEY §2.12.  The eager reference path materializes fp32
// log-softmax of [B*L, V] (measured 6.5 ms/step at V=27278 plus ~3 ms of
// bf16<->fp32 copies); here the loss reads the bf16 logits once
// (fused max+logsumexp, one wave per row, vectorized) and the backward
// writes bf16 dlogits in a second single pass: 3 passes over the logits
// instead of ~8, no fp32 copy ever.
//
// Layout: logits [N, V] bf16/fp32 (row-contiguous), labels [N] int64 with
// ignore_index for padded positions.  Loss = mean over valid rows.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// one 64-lane wave per row; 256-thread workgroups = 4 rows per block
template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              float* __restrict__ lse_out,
                              float* __restrict__ loss_out,  // scalar, atomic
                              int* __restrict__ count_out,   // scalar, atomic
                              int64_t n_rows, int64_t n_cols, int64_t ignore_index) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = (gridDim.x * blockDim.x) / WAVE;

  float local_loss = 0.f;
  int local_count = 0;
  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const int64_t label = labels[row];
    const T* lr = logits + row * n_cols;
    // pass 1: row max (vectorized 4-wide when aligned)
    float m = -INFINITY;
    int64_t c = lane * 4;
    if ((n_cols & 3) == 0) {
      for (; c + 3 < n_cols; c += WAVE * 4) {
        float4 v4;
        const T* p = lr + c;
        v4.x = to_f32<T>(p[0]); v4.y = to_f32<T>(p[1]);
        v4.z = to_f32<T>(p[2]); v4.w = to_f32<T>(p[3]);
        m = fmaxf(m, fmaxf(fmaxf(v4.x, v4.y), fmaxf(v4.z, v4.w)));
      }
    } else {
      for (int64_t cc = lane; cc < n_cols; cc += WAVE) m = fmaxf(m, to_f32<T>(lr[cc]));
    }
    m = wave_reduce_max(m);
    // pass 2: sum exp
    float s = 0.f;
    if ((n_cols & 3) == 0) {
      for (c = lane * 4; c + 3 < n_cols; c += WAVE * 4) {
        const T* p = lr + c;
        s += __expf(to_f32<T>(p[0]) - m) + __expf(to_f32<T>(p[1]) - m) +
             __expf(to_f32<T>(p[2]) - m) + __expf(to_f32<T>(p[3]) - m);
      }
    } else {
      for (int64_t cc = lane; cc < n_cols; cc += WAVE) s += __expf(to_f32<T>(lr[cc]) - m);
    }
    s = wave_reduce_sum(s);
    const float lse = m + __logf(s);
    if (lane == 0) {
      lse_out[row] = lse;
      if (label != ignore_index) {
        local_loss += lse - to_f32<T>(lr[label]);
        local_count += 1;
      }
    }
  }
  if (lane == 0) {
    if (local_count > 0) {
      atomicAdd(loss_out, local_loss);
      atomicAdd(count_out, local_count);
    }
  }
}

// dlogits[r, c] = (exp(logit - lse) - 1{c==label}) * scale  (scale = dloss/N)
template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              const float* __restrict__ lse,
                              T* __restrict__ dlogits,
                              const float* __restrict__ grad_scale,  // dLoss (scalar tensor)
                              const int* __restrict__ count,
                              int64_t n_rows, int64_t n_cols, int64_t ignore_index) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = (gridDim.x * blockDim.x) / WAVE;
  const float scale = grad_scale[0] / max(1, count[0]);

  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const int64_t label = labels[row];
    const T* lr = logits + row * n_cols;
    T* dr = dlogits + row * n_cols;
    if (label == ignore_index) {
      for (int64_t c = lane; c < n_cols; c += WAVE) dr[c] = from_f32<T>(0.f);
      continue;
    }
    const float l = lse[row];
    for (int64_t c = lane; c < n_cols; c += WAVE) {
      float p = __expf(to_f32<T>(lr[c]) - l);
      float g = (p - (c == label ? 1.f : 0.f)) * scale;
      dr[c] = from_f32<T>(g);
    }
  }
}

}  // namespace

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                  int64_t ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(labels.is_cuda() && labels.scalar_type() == torch::kLong);
  const int64_t n_rows = logits.size(0), n_cols = logits.size(1);
  auto f32 = logits.options().dtype(torch::kFloat32);
  auto lse = torch::empty({n_rows}, f32);
  auto loss = torch::zeros({1}, f32);
  auto count = torch::zeros({1}, logits.options().dtype(torch::kInt32));
  const int threads = 256;
  int blocks = (int)std::min<int64_t>((n_rows + 3) / 4, 4096);
  auto stream = at::cuda::getCurrentHIPStream();
  auto lab = labels.contiguous();
#define LAUNCH_CE_FWD(T)                                                              \
  hipLaunchKernelGGL(ce_fwd_kernel<T>, dim3(blocks), dim3(threads), 0, stream,        \
                     reinterpret_cast<const T*>(logits.data_ptr()),                   \
                     lab.data_ptr<int64_t>(), lse.data_ptr<float>(),                  \
                     loss.data_ptr<float>(), count.data_ptr<int>(), n_rows, n_cols,   \
                     ignore_index)
  if (logits.scalar_type() == torch::kBFloat16) {
    LAUNCH_CE_FWD(__hip_bfloat16);
  } else if (logits.scalar_type() == torch::kFloat32) {
    LAUNCH_CE_FWD(float);
  } else if (logits.scalar_type() == torch::kHalf) {
    LAUNCH_CE_FWD(__half);
  } else {
    TORCH_CHECK(false, "unsupported dtype");
  }
#undef LAUNCH_CE_FWD
  return {loss, count, lse};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
                     torch::Tensor grad_scale, torch::Tensor count, int64_t ignore_index,
                     bool inplace) {
  const int64_t n_rows = logits.size(0), n_cols = logits.size(1);
  auto dlogits = inplace ? logits : torch::empty_like(logits);
  const int threads = 256;
  int blocks = (int)std::min<int64_t>((n_rows + 3) / 4, 4096);
  auto stream = at::cuda::getCurrentHIPStream();
  auto lab = labels.contiguous();
  auto gs = grad_scale.to(torch::kFloat32).contiguous();
#define LAUNCH_CE_BWD(T)                                                              \
  hipLaunchKernelGGL(ce_bwd_kernel<T>, dim3(blocks), dim3(threads), 0, stream,        \
                     reinterpret_cast<const T*>(logits.data_ptr()),                   \
                     lab.data_ptr<int64_t>(), lse.data_ptr<float>(),                  \
                     reinterpret_cast<T*>(dlogits.data_ptr()), gs.data_ptr<float>(),  \
                     count.data_ptr<int>(), n_rows, n_cols, ignore_index)
  if (logits.scalar_type() == torch::kBFloat16) {
    LAUNCH_CE_BWD(__hip_bfloat16);
  } else if (logits.scalar_type() == torch::kFloat32) {
    LAUNCH_CE_BWD(float);
  } else if (logits.scalar_type() == torch::kHalf) {
    LAUNCH_CE_BWD(__half);
  } else {
    TORCH_CHECK(false, "unsupported dtype");
  }
#undef LAUNCH_CE_BWD
  return dlogits;
}
