// MFMA attention forward for gfx950 (CDNA4) — bf16, head_dim 32/64, L <= 256.
//
// The matrix-core version of K1/K2 (attention.hip keeps the VALU fallback
// for other shapes and the backward).  One workgroup = one (batch, head);
// each of the 4 waves owns a 16-query row tile and walks 16-key tiles:
//
//   S-tile [16q x 16k] = v_mfma_f32_16x16x32_bf16 over head_dim
//     A = Q fragment (resident in VGPRs, loaded once),
//     B = K fragment (16-B contiguous loads straight from row-major K),
//   causal/padding mask + online softmax on the accumulators (row groups =
//     16-lane shfl reductions), P converted bf16 and bounced through a
//     PER-WAVE LDS tile (wave-synchronous: no barriers) to re-enter the
//     MFMA pipe in A-fragment layout,
//   O-tile [16q x Dh] += P @ V via MFMA with V TRANSPOSED once into LDS at
//     workgroup start (B-fragments then read 16-B contiguous key runs).
//
// Numerics identical to the VALU kernel: fp32 accumulation, exact online
// softmax (no defer-max), the DefaultAttentionMask diagonal-rescue rule.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// C-layout row groups: row = (l>>4)*4 + r, col = l&15 -> the 16 lanes of a
// row are consecutive, i.e. one DPP "row".  The reduction runs on DPP
// row_ror permutes (pure VALU, 0x120|N ctrl): the __shfl_xor form
// compiled to ds_bpermute + a full lgkmcnt(0) drain per step — ~32
// serialized LDS round trips per key tile in the online softmax
// (measured dominant stall in the .s).  ror by 8/4/2/1 leaves every
// lane holding the full 16-lane reduction, same as the xor ladder.
#define ROW_ROR_DPP(x, N)   __builtin_amdgcn_update_dpp(0, (x), 0x120 | (N), 0xF, 0xF, true)
__device__ __forceinline__ float group16_max(float v) {
  v = fmaxf(v, __int_as_float(ROW_ROR_DPP(__float_as_int(v), 8)));
  v = fmaxf(v, __int_as_float(ROW_ROR_DPP(__float_as_int(v), 4)));
  v = fmaxf(v, __int_as_float(ROW_ROR_DPP(__float_as_int(v), 2)));
  v = fmaxf(v, __int_as_float(ROW_ROR_DPP(__float_as_int(v), 1)));
  return v;
}

__device__ __forceinline__ float group16_sum(float v) {
  v += __int_as_float(ROW_ROR_DPP(__float_as_int(v), 8));
  v += __int_as_float(ROW_ROR_DPP(__float_as_int(v), 4));
  v += __int_as_float(ROW_ROR_DPP(__float_as_int(v), 2));
  v += __int_as_float(ROW_ROR_DPP(__float_as_int(v), 1));
  return v;
}

template <int DH>  // head_dim: 32 or 64
__global__ __launch_bounds__(256, 2) void attn_fwd_mfma_kernel(
    const __hip_bfloat16* __restrict__ q,  // [B, H, L, DH]
    const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const bool* __restrict__ valid,  // [B, L] or nullptr
    __hip_bfloat16* __restrict__ out,
    float* __restrict__ lse_out,  // [B, H, L] or nullptr
    int B, int H, int L, float scale, bool causal) {
  constexpr int KS = DH / 32;     // MFMAs per S-tile
  constexpr int OF = DH / 16;     // O column fragments
  const int bh = blockIdx.x;
  const int b = bh / H;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int Lpad = (L + 31) & ~31;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // V^T [DH][Lpad] bf16 (+swizzle), then 4 per-wave P tiles [16][16] bf16
  __hip_bfloat16* vt = reinterpret_cast<__hip_bfloat16*>(smem);
  __hip_bfloat16* p_tiles = vt + (size_t)DH * Lpad;
  unsigned char* val_s = reinterpret_cast<unsigned char*>(p_tiles + 4 * 16 * 32);

  const size_t base = (size_t)bh * L * DH;
  // ---- stage V transposed (+1-free: row stride Lpad*2 B; swizzle by dh&7) --
  auto vt_off = [&](int dh, int key) {
    return ((size_t)dh * Lpad + (size_t)key) * 2;  // byte offset, swizzled below
  };
  for (int i = threadIdx.x; i < Lpad * DH; i += blockDim.x) {
    const int key = i / DH, dh = i % DH;
    const size_t byte = vt_off(dh, key) ^ (((size_t)(dh & 7)) << 4);
    // keys >= L zero-filled: their P is 0 but 0 * LDS-garbage could be NaN
    const __hip_bfloat16 val = (key < L) ? v[base + (size_t)key * DH + dh] : __hip_bfloat16(0.f);
    *reinterpret_cast<__hip_bfloat16*>(reinterpret_cast<char*>(vt) + byte) = val;
  }
  for (int i = threadIdx.x; i < L; i += blockDim.x) {
    val_s[i] = valid ? (unsigned char)valid[(size_t)b * L + i] : 1;
  }
  __syncthreads();

  __hip_bfloat16* my_p = p_tiles + wave * 512;  // wave-private [16][32]

  const int n_qtiles = (L + 15) >> 4;
  const int n_ktiles32 = (Lpad + 31) >> 5;
  for (int qt = wave; qt < n_qtiles; qt += 4) {
    const int q0 = qt << 4;
    // ---- A fragments of this Q tile, resident ----
    bf16x8 a_frag[KS];
    {
      const int row = q0 + (lane & 15);
      const int kk0 = (lane >> 4) * 8;
      const __hip_bfloat16* qr = q + base + (size_t)min(row, L - 1) * DH;
#pragma unroll
      for (int s = 0; s < KS; ++s) {
        a_frag[s] = *reinterpret_cast<const bf16x8*>(qr + s * 32 + kk0);
      }
    }
    // online-softmax state: each lane tracks its 4 rows (r = 0..3)
    float m_run[4], l_run[4];
    f32x4 o_acc[OF];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[r] = -INFINITY;
      l_run[r] = 0.f;
    }
#pragma unroll
    for (int f = 0; f < OF; ++f) o_acc[f] = f32x4{0.f, 0.f, 0.f, 0.f};

    // 32-key tiles: P tile [16 x 32] is exactly one MFMA A operand for PV
    const int kt_end = causal ? ((q0 + 15) >> 5) + 1 : n_ktiles32;
    for (int kt = 0; kt < kt_end; ++kt) {
      const int k0 = kt << 5;
      // ---- S tile = Q . K^T : two 16-col fragments ----
      f32x4 s_acc[2] = {f32x4{0.f, 0.f, 0.f, 0.f}, f32x4{0.f, 0.f, 0.f, 0.f}};
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int key = k0 + half * 16 + (lane & 15);
        const int kk0 = (lane >> 4) * 8;
        const __hip_bfloat16* kr = k + base + (size_t)min(key, L - 1) * DH;
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(kr + s * 32 + kk0);
          s_acc[half] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[s], b_frag, s_acc[half], 0, 0, 0);
        }
      }
      // ---- mask + online softmax on accumulators ----
      float p_val[2][4];
      float alpha[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = q0 + (lane >> 4) * 4 + r;
        float s_v[2];
#pragma unroll
        for (int half = 0; half < 2; ++half) {
          const int col = k0 + half * 16 + (lane & 15);
          bool allowed = (col == row) ||
                         ((!causal || col <= row) && col < L && val_s[min(col, L - 1)]);
          if (row >= L) allowed = (col == row);  // pad rows: diagonal only
          s_v[half] = allowed ? s_acc[half][r] * scale : -INFINITY;
        }
        float m_new = fmaxf(m_run[r], group16_max(fmaxf(s_v[0], s_v[1])));
        alpha[r] = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - m_new);
#pragma unroll
        for (int half = 0; half < 2; ++half) {
          p_val[half][r] = (s_v[half] == -INFINITY) ? 0.f : __expf(s_v[half] - m_new);
        }
        l_run[r] = l_run[r] * alpha[r] + group16_sum(p_val[0][r] + p_val[1][r]);
        m_run[r] = m_new;
      }
      // ---- P -> wave-private LDS [16][32] (C layout in, A layout out) ----
#pragma unroll
      for (int half = 0; half < 2; ++half) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          my_p[((lane >> 4) * 4 + r) * 32 + half * 16 + (lane & 15)] =
              __float2bfloat16(p_val[half][r]);
        }
      }
      // rescale O by alpha (per row r)
#pragma unroll
      for (int f = 0; f < OF; ++f) {
#pragma unroll
        for (int r = 0; r < 4; ++r) o_acc[f][r] *= alpha[r];
      }
      // (hipcc inserts the lgkmcnt wait between the LDS writes and reads;
      // wave-synchronous, no barrier needed)
      // ---- O += P @ V ----
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(my_p + (lane & 15) * 32 + (lane >> 4) * 8);
#pragma unroll
      for (int f = 0; f < OF; ++f) {
        // B = V^T fragment: lane holds V^T[dh = f*16 + l&15][keys (l>>4)*8..+7]
        const int dh = f * 16 + (lane & 15);
        const size_t byte = (vt_off(dh, k0 + (lane >> 4) * 8)) ^ (((size_t)(dh & 7)) << 4);
        bf16x8 vb = *reinterpret_cast<const bf16x8*>(reinterpret_cast<const char*>(vt) + byte);
        o_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb, o_acc[f], 0, 0, 0);
      }
    }
    // ---- epilogue: normalize + store ----
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = q0 + (lane >> 4) * 4 + r;
      if (row < L && lse_out != nullptr && (lane & 15) == 0) {
        lse_out[(size_t)bh * L + row] = m_run[r] + __logf(fmaxf(l_run[r], 1e-30f));
      }
    }
#pragma unroll
    for (int f = 0; f < OF; ++f) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = q0 + (lane >> 4) * 4 + r;
        if (row < L) {
          const float inv = 1.f / fmaxf(l_run[r], 1e-30f);
          out[base + (size_t)row * DH + f * 16 + (lane & 15)] =
              __float2bfloat16(o_acc[f][r] * inv);
        }
      }
    }
  }
}

}  // namespace

std::vector<torch::Tensor> attention_fwd_mfma(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                              c10::optional<torch::Tensor> valid, double scale,
                                              bool causal, bool need_lse) {
  TORCH_CHECK(q.dim() == 4 && q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16, "MFMA attention is bf16");
  const int B = q.size(0), H = q.size(1), L = q.size(2), D = q.size(3);
  TORCH_CHECK((D == 32 || D == 64) && L <= 256, "MFMA attention: D in {32,64}, L<=256");
  auto out = torch::empty_like(q);
  auto lse = torch::empty({B, H, L}, q.options().dtype(torch::kFloat32));
  const int Lpad = (L + 31) & ~31;
  size_t lds = (size_t)D * Lpad * 2 + 4 * 512 * 2 + L + 64;
  lds = (lds + 15) & ~size_t(15);
  auto stream = at::cuda::getCurrentHIPStream();
  const bool* valid_ptr = nullptr;
  torch::Tensor valid_c;
  if (valid.has_value()) {
    valid_c = valid->contiguous();
    valid_ptr = valid_c.data_ptr<bool>();
  }
#define LAUNCH_AF_MFMA(DD)                                                              \
  hipLaunchKernelGGL((attn_fwd_mfma_kernel<DD>), dim3(B * H), dim3(256), lds, stream,   \
                     reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),             \
                     reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),             \
                     reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()), valid_ptr,  \
                     reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),                 \
                     need_lse ? lse.data_ptr<float>() : nullptr, B, H, L, (float)scale, \
                     causal)
  if (D == 32) {
    LAUNCH_AF_MFMA(32);
  } else {
    LAUNCH_AF_MFMA(64);
  }
#undef LAUNCH_AF_MFMA
  return {out, lse};
}
