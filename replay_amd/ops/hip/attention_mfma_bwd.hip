// MFMA attention backward for gfx950 (CDNA4) — bf16, head_dim 32/64, L <= 256.
//
// Flash-style recompute backward on the matrix cores, two phases inside one
// workgroup per (batch, head) so no cross-workgroup accumulation is needed:
//
//   Phase A (waves own 16-QUERY tiles, walk 32-key tiles):
//     S = Q.K^T, P = exp(S*scale - lse)  (lse saved by the forward),
//     dP = dO.V^T, dS = P*(dP - delta[q])*scale,
//     dQ += dS.K  (dS bounced through wave-private LDS into A layout;
//                  K^T staged in LDS so B-fragments read contiguous).
//   Phase B (waves own 16-KEY tiles, walk 32-query tiles):
//     S^T = K.Q^T and dP^T = V.dO^T computed DIRECTLY (operand swap keeps
//     every B-fragment a contiguous 16-byte run), P^T/dS^T elementwise with
//     delta[col], then dK += dS^T.Q and dV += P^T.dO via LDS-staged Q^T/dO^T.
//
//   delta[q] = sum_d dO[q,d]*O[q,d] computed in the prologue.
//
// Replaces the VALU backward (attention.hip) for bf16 D in {32, 64}; that
// kernel remains the fallback and the numerics cross-check.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ float bwd_group16_sum(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

template <int DH>
__global__ __launch_bounds__(256, 2) void attn_bwd_mfma_kernel(
    const __hip_bfloat16* __restrict__ q,   // [B, H, L, DH]
    const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const __hip_bfloat16* __restrict__ out,
    const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse,   // [B, H, L]
    const bool* __restrict__ valid,  // [B, L] or nullptr
    __hip_bfloat16* __restrict__ dq,
    __hip_bfloat16* __restrict__ dk,
    __hip_bfloat16* __restrict__ dv,
    int B, int H, int L, float scale, bool causal) {
  constexpr int KS = DH / 32;
  constexpr int OF = DH / 16;
  const int bh = blockIdx.x;
  const int b = bh / H;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int Lpad = (L + 31) & ~31;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // transposed [DH][Lpad] stages: K^T (phase A), Q^T and dO^T (phase B)
  __hip_bfloat16* kt = reinterpret_cast<__hip_bfloat16*>(smem);
  __hip_bfloat16* qt = kt + (size_t)DH * Lpad;
  __hip_bfloat16* dot_s = qt + (size_t)DH * Lpad;
  float* delta_s = reinterpret_cast<float*>(dot_s + (size_t)DH * Lpad);  // Lpad
  __hip_bfloat16* p_tiles = reinterpret_cast<__hip_bfloat16*>(delta_s + Lpad);  // 4 x [16][32]
  __hip_bfloat16* ds_tiles = p_tiles + 4 * 512;                                  // 4 x [16][32]
  unsigned char* val_s = reinterpret_cast<unsigned char*>(ds_tiles + 4 * 512);

  const size_t base = (size_t)bh * L * DH;
  auto tr_off = [&](int dh, int pos) {
    return (((size_t)dh * Lpad + (size_t)pos) * 2) ^ (((size_t)(dh & 7)) << 4);
  };
  for (int i = threadIdx.x; i < Lpad * DH; i += blockDim.x) {
    const int pos = i / DH, dh = i % DH;
    const bool in_range = pos < L;
    const size_t src = base + (size_t)pos * DH + dh;
    const __hip_bfloat16 zero = __hip_bfloat16(0.f);
    const size_t byte = tr_off(dh, pos);
    *reinterpret_cast<__hip_bfloat16*>(reinterpret_cast<char*>(kt) + byte) =
        in_range ? k[src] : zero;
    *reinterpret_cast<__hip_bfloat16*>(reinterpret_cast<char*>(qt) + byte) =
        in_range ? q[src] : zero;
    *reinterpret_cast<__hip_bfloat16*>(reinterpret_cast<char*>(dot_s) + byte) =
        in_range ? dout[src] : zero;
  }
  for (int i = threadIdx.x; i < L; i += blockDim.x) {
    val_s[i] = valid ? (unsigned char)valid[(size_t)b * L + i] : 1;
  }
  // delta[q] = sum_d dO.O; one wave per row stripe
  for (int row = wave; row < Lpad; row += 4) {
    float acc = 0.f;
    if (row < L) {
      for (int d = lane; d < DH; d += WAVE) {
        acc += __bfloat162float(dout[base + (size_t)row * DH + d]) *
               __bfloat162float(out[base + (size_t)row * DH + d]);
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_xor(acc, off, WAVE);
    if (lane == 0) delta_s[row] = acc;
  }
  __syncthreads();

  __hip_bfloat16* my_p = p_tiles + wave * 512;
  __hip_bfloat16* my_ds = ds_tiles + wave * 512;
  const int n_t16 = Lpad >> 4;
  const int n_t32 = Lpad >> 5;

  // ======================= Phase A: dQ (q-owner) ==========================
  for (int qt16 = wave; qt16 < n_t16; qt16 += 4) {
    const int q0 = qt16 << 4;
    bf16x8 a_q[KS], a_do[KS];
    {
      const int row = q0 + (lane & 15);
      const int kk0 = (lane >> 4) * 8;
      const size_t rbase = base + (size_t)min(row, L - 1) * DH;
#pragma unroll
      for (int s = 0; s < KS; ++s) {
        a_q[s] = *reinterpret_cast<const bf16x8*>(q + rbase + s * 32 + kk0);
        a_do[s] = *reinterpret_cast<const bf16x8*>(dout + rbase + s * 32 + kk0);
      }
    }
    float my_lse[4], my_delta[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = q0 + (lane >> 4) * 4 + r;
      my_lse[r] = (row < L) ? lse[(size_t)bh * L + row] : 0.f;
      my_delta[r] = (row < L) ? delta_s[row] : 0.f;
    }
    f32x4 dq_acc[OF];
#pragma unroll
    for (int f = 0; f < OF; ++f) dq_acc[f] = f32x4{0.f, 0.f, 0.f, 0.f};

    const int kt_end = causal ? ((q0 + 15) >> 5) + 1 : n_t32;
    // K/V prefetched one key-tile ahead in registers: the direct load->MFMA
    // form compiled to a full vmcnt(0) drain per use (~10 per tile,
    // measured 80% wave parking); with the ring the waits are counted and
    // overlap the previous tile's epilogue.
    auto load_kv = [&](bf16x8 (&bk)[2][KS], bf16x8 (&bv)[2][KS], int kt32) {
      const int kk0 = (lane >> 4) * 8;
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int key = (kt32 << 5) + half * 16 + (lane & 15);
        const size_t kbase = base + (size_t)min(key, L - 1) * DH;
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          bk[half][s] = *reinterpret_cast<const bf16x8*>(k + kbase + s * 32 + kk0);
          bv[half][s] = *reinterpret_cast<const bf16x8*>(v + kbase + s * 32 + kk0);
        }
      }
    };
    bf16x8 bk_cur[2][KS], bv_cur[2][KS], bk_nxt[2][KS], bv_nxt[2][KS];
    load_kv(bk_cur, bv_cur, 0);
    for (int ktile = 0; ktile < kt_end; ++ktile) {  // NOT "kt": shadows the LDS stage ptr
      const int k0 = ktile << 5;
      if (ktile + 1 < kt_end) load_kv(bk_nxt, bv_nxt, ktile + 1);
      f32x4 s_acc[2] = {f32x4{0.f, 0.f, 0.f, 0.f}, f32x4{0.f, 0.f, 0.f, 0.f}};
      f32x4 dp_acc[2] = {f32x4{0.f, 0.f, 0.f, 0.f}, f32x4{0.f, 0.f, 0.f, 0.f}};
#pragma unroll
      for (int half = 0; half < 2; ++half) {
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          s_acc[half] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q[s], bk_cur[half][s], s_acc[half], 0, 0, 0);
          dp_acc[half] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_do[s], bv_cur[half][s],
                                                                 dp_acc[half], 0, 0, 0);
        }
      }
      // dS = P * (dP - delta) * scale, written to my_ds [16][32]
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int col = k0 + half * 16 + (lane & 15);
        // r-invariant hoists: the in-loop val_s byte read compiled to a
        // ds_read_u8 + lgkmcnt(0) per (half, r)
        const bool vcol = (col < L) && val_s[min(col, L - 1)];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = q0 + (lane >> 4) * 4 + r;
          bool allowed = (col == row) || ((!causal || col <= row) && vcol);
          if (row >= L) allowed = false;
          float p = allowed ? __expf(s_acc[half][r] * scale - my_lse[r]) : 0.f;
          float ds_v = p * (dp_acc[half][r] - my_delta[r]) * scale;
          my_ds[((lane >> 4) * 4 + r) * 32 + half * 16 + (lane & 15)] = __float2bfloat16(ds_v);
        }
      }
      // dQ += dS . K  (A = bounced dS, B = K^T fragments from LDS)
      bf16x8 dsa = *reinterpret_cast<const bf16x8*>(my_ds + (lane & 15) * 32 + (lane >> 4) * 8);
#pragma unroll
      for (int f = 0; f < OF; ++f) {
        const int dh = f * 16 + (lane & 15);
        const size_t byte = tr_off(dh, k0 + (lane >> 4) * 8);
        bf16x8 kb = *reinterpret_cast<const bf16x8*>(reinterpret_cast<const char*>(kt) + byte);
        dq_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa, kb, dq_acc[f], 0, 0, 0);
      }
#pragma unroll
      for (int half = 0; half < 2; ++half) {
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          bk_cur[half][s] = bk_nxt[half][s];
          bv_cur[half][s] = bv_nxt[half][s];
        }
      }
    }
#pragma unroll
    for (int f = 0; f < OF; ++f) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = q0 + (lane >> 4) * 4 + r;
        if (row < L) {
          dq[base + (size_t)row * DH + f * 16 + (lane & 15)] = __float2bfloat16(dq_acc[f][r]);
        }
      }
    }
  }

  // ======================= Phase B: dK, dV (k-owner) ======================
  for (int kt16 = wave; kt16 < n_t16; kt16 += 4) {
    const int k0 = kt16 << 4;
    bf16x8 a_k[KS], a_v[KS];
    {
      const int key = k0 + (lane & 15);
      const int kk0 = (lane >> 4) * 8;
      const size_t kbase = base + (size_t)min(key, L - 1) * DH;
#pragma unroll
      for (int s = 0; s < KS; ++s) {
        a_k[s] = *reinterpret_cast<const bf16x8*>(k + kbase + s * 32 + kk0);
        a_v[s] = *reinterpret_cast<const bf16x8*>(v + kbase + s * 32 + kk0);
      }
    }
    f32x4 dk_acc[OF], dv_acc[OF];
#pragma unroll
    for (int f = 0; f < OF; ++f) {
      dk_acc[f] = f32x4{0.f, 0.f, 0.f, 0.f};
      dv_acc[f] = f32x4{0.f, 0.f, 0.f, 0.f};
    }
    // qtile-INVARIANT per-key validity, hoisted: the in-loop byte read
    // compiled to ds_read_u8 + lgkmcnt(0) at every (half, r)
    bool vrow[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int krow = k0 + (lane >> 4) * 4 + r;
      vrow[r] = (krow < L) && val_s[min(krow, L - 1)];
    }
    // causal: queries >= k0 contribute; start at the 32-tile containing k0
    const int qt_start = causal ? (k0 >> 5) : 0;
    auto load_qdo = [&](bf16x8 (&bq)[2][KS], bf16x8 (&bdo)[2][KS], int qt32) {
      const int kk0 = (lane >> 4) * 8;
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int qrow = (qt32 << 5) + half * 16 + (lane & 15);
        const size_t qbase = base + (size_t)min(qrow, L - 1) * DH;
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          bq[half][s] = *reinterpret_cast<const bf16x8*>(q + qbase + s * 32 + kk0);
          bdo[half][s] = *reinterpret_cast<const bf16x8*>(dout + qbase + s * 32 + kk0);
        }
      }
    };
    bf16x8 bq_cur[2][KS], bdo_cur[2][KS], bq_nxt[2][KS], bdo_nxt[2][KS];
    if (qt_start < n_t32) load_qdo(bq_cur, bdo_cur, qt_start);
    for (int qtile = qt_start; qtile < n_t32; ++qtile) {
      const int qq0 = qtile << 5;
      if (qtile + 1 < n_t32) load_qdo(bq_nxt, bdo_nxt, qtile + 1);
      // clamped unconditional lse/delta loads issued BEFORE the MFMA block
      // so their waits hide under it (the exec-masked conditional form
      // waited vmcnt(0) at first use)
      float lq2[2], dq2[2];
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int qcol = qq0 + half * 16 + (lane & 15);
        lq2[half] = lse[(size_t)bh * L + min(qcol, L - 1)];
        dq2[half] = delta_s[min(qcol, L - 1)];
      }
      f32x4 st_acc[2] = {f32x4{0.f, 0.f, 0.f, 0.f}, f32x4{0.f, 0.f, 0.f, 0.f}};
      f32x4 dpt_acc[2] = {f32x4{0.f, 0.f, 0.f, 0.f}, f32x4{0.f, 0.f, 0.f, 0.f}};
#pragma unroll
      for (int half = 0; half < 2; ++half) {
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          // S^T[k, q] = K . Q^T ; dP^T[k, q] = V . dO^T
          st_acc[half] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_k[s], bq_cur[half][s], st_acc[half], 0, 0, 0);
          dpt_acc[half] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_v[s], bdo_cur[half][s],
                                                                  dpt_acc[half], 0, 0, 0);
        }
      }
      // P^T and dS^T elementwise (rows = keys, cols = queries); the per-query
      // lse/delta are r-invariant — hoisted out of the r loop (a per-(r)
      // global lse gather serialized 4 redundant loads per half)
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int qcol = qq0 + half * 16 + (lane & 15);
        const float l_q = (qcol < L) ? lq2[half] : 0.f;
        const float d_q = (qcol < L) ? dq2[half] : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int krow = k0 + (lane >> 4) * 4 + r;
          bool allowed = (qcol == krow) || ((!causal || krow <= qcol) && vrow[r]);
          if (qcol >= L) allowed = false;
          float p = allowed ? __expf(st_acc[half][r] * scale - l_q) : 0.f;
          float ds_v = p * (dpt_acc[half][r] - d_q) * scale;
          const int idx = ((lane >> 4) * 4 + r) * 32 + half * 16 + (lane & 15);
          my_p[idx] = __float2bfloat16(p);
          my_ds[idx] = __float2bfloat16(ds_v);
        }
      }
      // dK += dS^T . Q ; dV += P^T . dO  (A = bounced tiles, B = LDS stages)
      bf16x8 dsa = *reinterpret_cast<const bf16x8*>(my_ds + (lane & 15) * 32 + (lane >> 4) * 8);
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(my_p + (lane & 15) * 32 + (lane >> 4) * 8);
#pragma unroll
      for (int f = 0; f < OF; ++f) {
        const int dh = f * 16 + (lane & 15);
        const size_t byte = tr_off(dh, qq0 + (lane >> 4) * 8);
        bf16x8 qb = *reinterpret_cast<const bf16x8*>(reinterpret_cast<const char*>(qt) + byte);
        bf16x8 dob =
            *reinterpret_cast<const bf16x8*>(reinterpret_cast<const char*>(dot_s) + byte);
        dk_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa, qb, dk_acc[f], 0, 0, 0);
        dv_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, dob, dv_acc[f], 0, 0, 0);
      }
#pragma unroll
      for (int half = 0; half < 2; ++half) {
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          bq_cur[half][s] = bq_nxt[half][s];
          bdo_cur[half][s] = bdo_nxt[half][s];
        }
      }
    }
#pragma unroll
    for (int f = 0; f < OF; ++f) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int krow = k0 + (lane >> 4) * 4 + r;
        if (krow < L) {
          dk[base + (size_t)krow * DH + f * 16 + (lane & 15)] = __float2bfloat16(dk_acc[f][r]);
          dv[base + (size_t)krow * DH + f * 16 + (lane & 15)] = __float2bfloat16(dv_acc[f][r]);
        }
      }
    }
  }
}

}  // namespace

std::vector<torch::Tensor> attention_bwd_mfma(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                              torch::Tensor out, torch::Tensor dout,
                                              torch::Tensor lse,
                                              c10::optional<torch::Tensor> valid, double scale,
                                              bool causal) {
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16);
  const int B = q.size(0), H = q.size(1), L = q.size(2), D = q.size(3);
  TORCH_CHECK((D == 32 || D == 64) && L <= 256);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  const int Lpad = (L + 31) & ~31;
  size_t lds = 3 * (size_t)D * Lpad * 2 + (size_t)Lpad * 4 + 8 * 512 * 2 + L + 64;
  lds = (lds + 15) & ~size_t(15);
  TORCH_CHECK(lds <= 160 * 1024, "LDS overflow");
  auto stream = at::cuda::getCurrentHIPStream();
  const bool* valid_ptr = nullptr;
  torch::Tensor valid_c;
  if (valid.has_value()) {
    valid_c = valid->contiguous();
    valid_ptr = valid_c.data_ptr<bool>();
  }
  auto dout_c = dout.contiguous();
#define LAUNCH_AB_MFMA(DD)                                                                \
  hipLaunchKernelGGL((attn_bwd_mfma_kernel<DD>), dim3(B * H), dim3(256), lds, stream,     \
                     reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),               \
                     reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),               \
                     reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),               \
                     reinterpret_cast<const __hip_bfloat16*>(out.data_ptr()),             \
                     reinterpret_cast<const __hip_bfloat16*>(dout_c.data_ptr()),          \
                     lse.data_ptr<float>(), valid_ptr,                                    \
                     reinterpret_cast<__hip_bfloat16*>(dq.data_ptr()),                    \
                     reinterpret_cast<__hip_bfloat16*>(dk.data_ptr()),                    \
                     reinterpret_cast<__hip_bfloat16*>(dv.data_ptr()), B, H, L,           \
                     (float)scale, causal)
  if (D == 32) {
    LAUNCH_AB_MFMA(32);
  } else {
    LAUNCH_AB_MFMA(64);
  }
#undef LAUNCH_AB_MFMA
  return {dq, dk, dv};
}
