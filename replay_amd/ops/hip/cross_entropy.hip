// Fused softmax cross-entropy over large catalogs for gfx950 (CDNA4).
//
// K10 in SURVEY §2.12.  One 64-lane wave per row of [N, V] logits:
// forward = ONE vectorized pass (online max + rescaled sum-exp, 8 bf16 =
// 16 B per lane per iteration, guide G13), backward = one vectorized pass
// writing bf16 dlogits in place.  Used standalone and as the per-chunk core
// of the chunked CE (ops/fused_ce.py) whose chunks stay L3-resident.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// load 8 bf16 (16 B) as uint4
__device__ __forceinline__ void load8_bf16(const __hip_bfloat16* p, float* out) {
  const uint4 raw = *reinterpret_cast<const uint4*>(p);
  const unsigned w[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    __hip_bfloat162 pair = *reinterpret_cast<const __hip_bfloat162*>(&w[i]);
    out[2 * i] = __bfloat162float(pair.x);
    out[2 * i + 1] = __bfloat162float(pair.y);
  }
}

__device__ __forceinline__ void store8_bf16(__hip_bfloat16* p, const float* in) {
  uint4 raw;
  unsigned* w = reinterpret_cast<unsigned*>(&raw);
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    __hip_bfloat162 pair = __float22bfloat162_rn(float2{in[2 * i], in[2 * i + 1]});
    w[i] = *reinterpret_cast<const unsigned*>(&pair);
  }
  *reinterpret_cast<uint4*>(p) = raw;
}

// online (max, sum) combine across lanes
__device__ __forceinline__ void wave_reduce_online(float& m, float& s) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float m2 = __shfl_xor(m, off, WAVE);
    float s2 = __shfl_xor(s, off, WAVE);
    float mn = fmaxf(m, m2);
    // lanes that saw no elements carry (m=-inf, s=0): guard 0*exp(-inf-(-inf))
    float a = (s == 0.f) ? 0.f : s * __expf(m - mn);
    float b = (s2 == 0.f) ? 0.f : s2 * __expf(m2 - mn);
    s = a + b;
    m = mn;
  }
}

template <typename T, bool VEC8>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              float* __restrict__ lse_out,
                              float* __restrict__ loss_out,
                              int* __restrict__ count_out,
                              int64_t n_rows, int64_t n_cols, int64_t ignore_index) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = (gridDim.x * blockDim.x) / WAVE;

  float local_loss = 0.f;
  int local_count = 0;
  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const T* lr = logits + row * n_cols;
    float m = -INFINITY, s = 0.f;
    if constexpr (VEC8) {
      // per-row 16B alignment: scalar head, vec8 body, scalar tail
      const int head = (int)(((16 - (reinterpret_cast<uintptr_t>(lr) & 15)) & 15) >> 1);
      const int64_t n_vec = head + ((n_cols - head) & ~int64_t(7));
      float v[8];
      for (int64_t c = lane; c < head; c += WAVE) {
        float x = to_f32<T>(lr[c]);
        if (x > m) { s *= __expf(m - x); m = x; }
        s += __expf(x - m);
      }
      int64_t c = head + lane * 8;
      // 2x unrolled with both loads issued before the transcendental work:
      // the single-chunk loop measured 84.7% SQ_WAIT_ANY (load-latency bound)
      for (; c + WAVE * 8 + 7 < n_vec; c += WAVE * 16) {
        float v2[8];
        load8_bf16(reinterpret_cast<const __hip_bfloat16*>(lr) + c, v);
        load8_bf16(reinterpret_cast<const __hip_bfloat16*>(lr) + c + WAVE * 8, v2);
        float m8 = v[0];
#pragma unroll
        for (int i = 1; i < 8; ++i) m8 = fmaxf(m8, v[i]);
#pragma unroll
        for (int i = 0; i < 8; ++i) m8 = fmaxf(m8, v2[i]);
        if (m8 > m) {
          s *= __expf(m - m8);
          m = m8;
        }
#pragma unroll
        for (int i = 0; i < 8; ++i) s += __expf(v[i] - m);
#pragma unroll
        for (int i = 0; i < 8; ++i) s += __expf(v2[i] - m);
      }
      for (; c + 7 < n_vec; c += WAVE * 8) {
        load8_bf16(reinterpret_cast<const __hip_bfloat16*>(lr) + c, v);
        float m8 = v[0];
#pragma unroll
        for (int i = 1; i < 8; ++i) m8 = fmaxf(m8, v[i]);
        if (m8 > m) {
          s *= __expf(m - m8);
          m = m8;
        }
#pragma unroll
        for (int i = 0; i < 8; ++i) s += __expf(v[i] - m);
      }
      for (int64_t c = n_vec + lane; c < n_cols; c += WAVE) {
        float x = to_f32<T>(lr[c]);
        if (x > m) { s *= __expf(m - x); m = x; }
        s += __expf(x - m);
      }
    } else {
      for (int64_t c = lane; c < n_cols; c += WAVE) {
        float v = to_f32<T>(lr[c]);
        if (v > m) {
          s *= __expf(m - v);
          m = v;
        }
        s += __expf(v - m);
      }
    }
    wave_reduce_online(m, s);
    const float lse = m + __logf(s);
    if (lane == 0) {
      lse_out[row] = lse;
      const int64_t label = labels[row];
      if (label != ignore_index) {
        local_loss += lse - to_f32<T>(lr[label]);
        local_count += 1;
      }
    }
  }
  if (lane == 0 && local_count > 0) {
    atomicAdd(loss_out, local_loss);
    atomicAdd(count_out, local_count);
  }
}

template <typename T, bool VEC8>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              const float* __restrict__ lse,
                              T* __restrict__ dlogits,
                              const float* __restrict__ grad_scale,
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
    const bool ignored = (label == ignore_index);
    const float l = ignored ? 0.f : lse[row];
    if constexpr (VEC8) {
      const int head = (int)(((16 - (reinterpret_cast<uintptr_t>(lr) & 15)) & 15) >> 1);
      const int64_t n_vec = head + ((n_cols - head) & ~int64_t(7));
      float v[8];
      for (int64_t c = lane; c < head; c += WAVE) {
        if (ignored) { dr[c] = from_f32<T>(0.f); }
        else {
          float p = __expf(to_f32<T>(lr[c]) - l);
          dr[c] = from_f32<T>((p - (c == label ? 1.f : 0.f)) * scale);
        }
      }
      int64_t c = head + lane * 8;
      for (; c + WAVE * 8 + 7 < n_vec; c += WAVE * 16) {
        float v2[8];
        if (ignored) {
#pragma unroll
          for (int i = 0; i < 8; ++i) v[i] = v2[i] = 0.f;
        } else {
          load8_bf16(reinterpret_cast<const __hip_bfloat16*>(lr) + c, v);
          load8_bf16(reinterpret_cast<const __hip_bfloat16*>(lr) + c + WAVE * 8, v2);
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            float p = __expf(v[i] - l);
            v[i] = (p - ((c + i) == label ? 1.f : 0.f)) * scale;
          }
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            float p = __expf(v2[i] - l);
            v2[i] = (p - ((c + WAVE * 8 + i) == label ? 1.f : 0.f)) * scale;
          }
        }
        store8_bf16(reinterpret_cast<__hip_bfloat16*>(dr) + c, v);
        store8_bf16(reinterpret_cast<__hip_bfloat16*>(dr) + c + WAVE * 8, v2);
      }
      for (; c + 7 < n_vec; c += WAVE * 8) {
        if (ignored) {
#pragma unroll
          for (int i = 0; i < 8; ++i) v[i] = 0.f;
        } else {
          load8_bf16(reinterpret_cast<const __hip_bfloat16*>(lr) + c, v);
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            float p = __expf(v[i] - l);
            v[i] = (p - ((c + i) == label ? 1.f : 0.f)) * scale;
          }
        }
        store8_bf16(reinterpret_cast<__hip_bfloat16*>(dr) + c, v);
      }
      for (int64_t c = n_vec + lane; c < n_cols; c += WAVE) {
        if (ignored) { dr[c] = from_f32<T>(0.f); }
        else {
          float p = __expf(to_f32<T>(lr[c]) - l);
          dr[c] = from_f32<T>((p - (c == label ? 1.f : 0.f)) * scale);
        }
      }
    } else {
      for (int64_t c = lane; c < n_cols; c += WAVE) {
        if (ignored) {
          dr[c] = from_f32<T>(0.f);
        } else {
          float p = __expf(to_f32<T>(lr[c]) - l);
          dr[c] = from_f32<T>((p - (c == label ? 1.f : 0.f)) * scale);
        }
      }
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
  // row alignment handled in-kernel (scalar head); vec8 for any bf16 shape
  const bool vec8 = (logits.scalar_type() == torch::kBFloat16);
#define LAUNCH_CE_FWD(T, V)                                                           \
  hipLaunchKernelGGL((ce_fwd_kernel<T, V>), dim3(blocks), dim3(threads), 0, stream,   \
                     reinterpret_cast<const T*>(logits.data_ptr()),                   \
                     lab.data_ptr<int64_t>(), lse.data_ptr<float>(),                  \
                     loss.data_ptr<float>(), count.data_ptr<int>(), n_rows, n_cols,   \
                     ignore_index)
  if (logits.scalar_type() == torch::kBFloat16) {
    if (vec8) LAUNCH_CE_FWD(__hip_bfloat16, true);
    else LAUNCH_CE_FWD(__hip_bfloat16, false);
  } else if (logits.scalar_type() == torch::kFloat32) {
    LAUNCH_CE_FWD(float, false);
  } else if (logits.scalar_type() == torch::kHalf) {
    LAUNCH_CE_FWD(__half, false);
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
  const bool vec8 = (logits.scalar_type() == torch::kBFloat16);
#define LAUNCH_CE_BWD(T, V)                                                           \
  hipLaunchKernelGGL((ce_bwd_kernel<T, V>), dim3(blocks), dim3(threads), 0, stream,   \
                     reinterpret_cast<const T*>(logits.data_ptr()),                   \
                     lab.data_ptr<int64_t>(), lse.data_ptr<float>(),                  \
                     reinterpret_cast<T*>(dlogits.data_ptr()), gs.data_ptr<float>(),  \
                     count.data_ptr<int>(), n_rows, n_cols, ignore_index)
  if (logits.scalar_type() == torch::kBFloat16) {
    if (vec8) LAUNCH_CE_BWD(__hip_bfloat16, true);
    else LAUNCH_CE_BWD(__hip_bfloat16, false);
  } else if (logits.scalar_type() == torch::kFloat32) {
    LAUNCH_CE_BWD(float, false);
  } else if (logits.scalar_type() == torch::kHalf) {
    LAUNCH_CE_BWD(__half, false);
  } else {
    TORCH_CHECK(false, "unsupported dtype");
  }
#undef LAUNCH_CE_BWD
  return dlogits;
}
