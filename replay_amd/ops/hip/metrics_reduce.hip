// Fused ranking-metric reduction for gfx950 (K13 in SURVEY §2.12).
//
// Replaces the TorchMetricsBuilder's per-k eager op chain (broadcast
// compare [B, K, G] + ~8 reduction launches per cutoff, reference
// replay/metrics/torch_metrics_builder.py:306-349) with ONE launch that
// computes the per-batch SUMS of hitrate/recall/precision/ndcg/map/mrr/
// novelty at every cutoff.  One thread owns one user row (the tensors are
// tiny — B x K<=~20 predictions, B x G ground truth); hits come from a
// register loop over the row's ground truth (L1-resident), per-counter
// partial sums are wave-reduced and land in global fp64 accumulators with
// one atomic per wave per counter.
//
// Metric definitions match the builder exactly (itself doctest-matched to
// the reference offline metrics): dcg weight 1/log2(pos+2), idcg over
// min(gt_count, k), ap normalized by min(gt_count, k), novelty = 1 - |hits
// in train|/k.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int N_METRICS = 7;  // hitrate, recall, precision, ndcg, map, mrr, novelty
constexpr int MAX_K = 64;
constexpr int MAX_CUTS = 8;

__device__ __forceinline__ double wave_sum_f64(double v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    v += __shfl_down(v, off, WAVE);
  }
  return v;
}

__global__ void metrics_reduce_kernel(
    const int64_t* __restrict__ preds,  // [B, K] best-first
    const int64_t* __restrict__ gt,     // [B, G], -1 padded
    const int64_t* __restrict__ train,  // [B, T], -1 padded (nullptr: no novelty)
    const int* __restrict__ ks,         // [nk] ascending cutoffs
    double* __restrict__ out,           // [N_METRICS, nk]
    int B, int K, int G, int T, int nk) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const bool active = row < B;

  float hits[MAX_K];
  float gt_count = 0.f;
  int kk = min(K, MAX_K);
  if (active) {
    const int64_t* gr = gt + (size_t)row * G;
    for (int g = 0; g < G; ++g) gt_count += (gr[g] >= 0) ? 1.f : 0.f;
    const int64_t* pr = preds + (size_t)row * K;
    for (int j = 0; j < kk; ++j) {
      const int64_t p = pr[j];
      float h = 0.f;
      for (int g = 0; g < G; ++g) {
        h = fmaxf(h, (gr[g] >= 0 && gr[g] == p) ? 1.f : 0.f);
      }
      hits[j] = h;
    }
  } else {
    for (int j = 0; j < kk; ++j) hits[j] = 0.f;
  }
  const float gt_c = fmaxf(gt_count, 1.f);

  // per-cutoff accumulators for this row
  double acc[N_METRICS][MAX_CUTS];
  for (int m = 0; m < N_METRICS; ++m)
    for (int c = 0; c < nk; ++c) acc[m][c] = 0.0;

  if (active) {
    float cum = 0.f, dcg = 0.f, ap = 0.f, rr = 0.f, seen_in_train = 0.f;
    bool any = false;
    int ci = 0;
    const int64_t* tr = (train != nullptr) ? train + (size_t)row * T : nullptr;
    const int64_t* pr = preds + (size_t)row * K;
    for (int j = 0; j < kk && ci < nk; ++j) {
      const float h = hits[j];
      cum += h;
      dcg += h / __log2f((float)j + 2.f);
      ap += h * (cum / (float)(j + 1));
      if (h > 0.f && !any) {
        rr = 1.f / (float)(j + 1);
        any = true;
      }
      if (tr != nullptr) {
        const int64_t p = pr[j];
        float s = 0.f;
        for (int t = 0; t < T; ++t) s = fmaxf(s, (tr[t] >= 0 && tr[t] == p) ? 1.f : 0.f);
        seen_in_train += s;
      }
      while (ci < nk && ks[ci] == j + 1) {
        const int k = ks[ci];
        const float ideal_n = fminf(gt_count, (float)k);
        float idcg = 0.f;
        for (int i = 0; i < (int)ideal_n; ++i) idcg += 1.f / __log2f((float)i + 2.f);
        acc[0][ci] = (cum > 0.f) ? 1.0 : 0.0;                        // hitrate
        acc[1][ci] = cum / gt_c;                                     // recall
        acc[2][ci] = cum / (float)k;                                 // precision
        acc[3][ci] = (idcg > 0.f) ? (double)(dcg / idcg) : 0.0;      // ndcg
        acc[4][ci] = (double)(ap / fmaxf(ideal_n, 1.f));             // map
        acc[5][ci] = (double)rr;                                     // mrr
        acc[6][ci] = (tr != nullptr) ? (double)(1.f - seen_in_train / (float)k) : 0.0;
        ++ci;
      }
    }
  }

  // wave-reduce each counter, one atomic per wave per counter
  for (int m = 0; m < N_METRICS; ++m) {
    for (int c = 0; c < nk; ++c) {
      const double s = wave_sum_f64(acc[m][c]);
      if (lane == 0 && s != 0.0) {
        atomicAdd(&out[m * nk + c], s);
      }
    }
  }
}

}  // namespace

torch::Tensor metrics_reduce(torch::Tensor preds, torch::Tensor gt,
                             c10::optional<torch::Tensor> train, torch::Tensor ks) {
  TORCH_CHECK(preds.is_cuda() && preds.dim() == 2 && preds.scalar_type() == torch::kLong);
  TORCH_CHECK(gt.is_cuda() && gt.dim() == 2 && gt.scalar_type() == torch::kLong);
  const int B = (int)preds.size(0);
  const int K = (int)preds.size(1);
  TORCH_CHECK(K <= 64, "metrics_reduce supports max_k <= 64");
  const int G = (int)gt.size(1);
  auto ks_c = ks.to(torch::kInt32).contiguous();
  const int nk = (int)ks_c.size(0);
  TORCH_CHECK(nk <= 8, "metrics_reduce supports <= 8 cutoffs");
  auto out = torch::zeros({7, nk}, preds.options().dtype(torch::kFloat64));
  const int64_t* train_ptr = nullptr;
  int T = 0;
  torch::Tensor train_c;
  if (train.has_value() && train->numel() > 0) {
    train_c = train->contiguous();
    train_ptr = train_c.data_ptr<int64_t>();
    T = (int)train_c.size(1);
  }
  auto preds_c = preds.contiguous();
  auto gt_c = gt.contiguous();
  auto stream = at::cuda::getCurrentHIPStream();
  const int threads = 256;
  const int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(metrics_reduce_kernel, dim3(blocks), dim3(threads), 0, stream,
                     preds_c.data_ptr<int64_t>(), gt_c.data_ptr<int64_t>(), train_ptr,
                     ks_c.data_ptr<int>(), out.data_ptr<double>(), B, K, G, T, nk);
  return out;
}
