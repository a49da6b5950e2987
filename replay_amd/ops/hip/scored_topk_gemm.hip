// Fused catalog-score GEMM + top-K candidate selection for gfx950 (CDNA4).
//
// K7+K8 in SURVEY §2.12 fully fused: scores[q, i] = Q[q,:] . W[i,:] computed
// on the MFMA matrix cores (v_mfma_f32_16x16x32_bf16) with the per-row
// threshold test applied IN THE EPILOGUE on the accumulator registers — the
// [B, V] score matrix never exists in memory.  At V = 10M items the unfused
// pipeline writes + re-reads 40 GB of scores per batch; this kernel's HBM
// traffic is the item table itself (V x E bf16, streamed once).
//
// Geometry: one workgroup = 4 waves = a 256-query M-tile; each wave owns 64
// query rows as 4 row-fragments whose A-fragments (64 x E bf16) stay
// RESIDENT in VGPRs for the whole launch (E = 256 -> 128 VGPRs/lane), and
// streams 64-item B-tiles with direct 16-B global loads shared by all 4
// row-fragments (16 MFMAs per 4 B loads per K-step).  The wide M-tile is
// the W-traffic lever: item-table bytes scale with ceil(M/256), not
// ceil(M/64).  Candidates (score >= per-query threshold) are rare by
// construction (threshold ~ kth value from a host-side subsample), so the
// epilogue's common path is 64 VALU compares per tile; hits append
// (value, index) to per-query buffers via one global atomicAdd each.
//
// Fragment maps (verified by the asymmetric-B GPU parity test, guide G9):
//   A (16x32 bf16): lane l holds A[row = l&15][k = (l>>4)*8 + j], j=0..7
//   B (32x16 bf16): lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C (16x16 f32):  lane l holds C[row = (l>>4)*4 + r][col = l&15], r=0..3

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"
#include <cstdlib>

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// RESIDENT: keep the full A (64 rows x E) in VGPRs across tiles (E <= 128);
// at E = 256 that spills (needs 128 VGPRs for A alone), so the WG's whole
// 256-row Q tile lives in LDS (128 KiB, XOR-swizzled so the 16-lane
// fragment-read groups are conflict-free, guide G4) and A-fragments are
// re-read per K-step at LDS speed.
template <int E, bool RESIDENT>
__global__ __launch_bounds__(256, 2) void scored_topk_gemm_kernel(
    const __hip_bfloat16* __restrict__ q,  // [M, E]
    const __hip_bfloat16* __restrict__ w,  // [V, E]
    const float* __restrict__ thresholds,  // [M]
    float* __restrict__ out_vals,          // [M, cap]
    int* __restrict__ out_idx,             // [M, cap]
    int* __restrict__ counts,              // [M]
    int M, int64_t V, int cap) {
  constexpr int KSTEPS = E / 32;
  constexpr int MF = 4;  // row-fragments per wave (wave covers 64 rows)
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int m0 = blockIdx.x * 256 + wave * 64;  // this wave's first query row

  // ---- A fragments (64 Q rows): resident across tiles when they fit ----
  constexpr int A_KS = RESIDENT ? KSTEPS : 1;
  bf16x8 a_frag[MF][A_KS];
  // non-resident path: the WG's 256xE Q tile in LDS, XOR-swizzled
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __hip_bfloat16* q_lds = reinterpret_cast<__hip_bfloat16*>(smem);
  auto lds_off = [&](int row, int k_byte) {
    return row * (E * 2) + (k_byte ^ ((row & 7) << 4));
  };
  if constexpr (!RESIDENT) {
    // cooperative staged load: thread t copies 16B pieces
    const int row_q0 = blockIdx.x * 256;
    for (int i = threadIdx.x; i < 256 * (E * 2 / 16); i += blockDim.x) {
      const int row = i / (E * 2 / 16);
      const int k_byte = (i % (E * 2 / 16)) * 16;
      const int src_row = min(row_q0 + row, M - 1);
      uint4 v = *reinterpret_cast<const uint4*>(
          reinterpret_cast<const char*>(q + (size_t)src_row * E) + k_byte);
      if (row_q0 + row >= M) v = uint4{0, 0, 0, 0};
      *reinterpret_cast<uint4*>(reinterpret_cast<char*>(q_lds) + lds_off(row, k_byte)) = v;
    }
    __syncthreads();
  }
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int row = m0 + mf * 16 + (lane & 15);
    const int k0 = (lane >> 4) * 8;
    if constexpr (RESIDENT) {
      const __hip_bfloat16* qr = q + (size_t)min(row, M - 1) * E + k0;
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        a_frag[mf][ks] = *reinterpret_cast<const bf16x8*>(qr + ks * 32);
      }
      if (row >= M) {
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks) a_frag[mf][ks] = bf16x8{0};
      }
    }
  }
  // per-lane thresholds of its MF x 4 C rows
  float t_reg[MF][4];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
      t_reg[mf][r] = (row < M) ? thresholds[row] : INFINITY;
    }
  }

  const int64_t n_tiles = (V + 63) >> 6;
  const int bk0 = (lane >> 4) * 8;
  // ---- software-pipelined B stream: depth-4 ring of 4-fragment groups ----
  // Each "group" g = (tile, ks) is 4 x 16 B loads; groups are prefetched
  // PIPE_D sections ahead (across tile boundaries) so ~1000 cycles of HBM
  // latency hide under the 16-MFMA sections (MFMA busy measured 9% without
  // this, waves parked in s_waitcnt 83%).
  constexpr int PIPE_D = RESIDENT ? 2 : 4;  // resident-A variants are register-tight
  static_assert(KSTEPS % PIPE_D == 0 || KSTEPS < PIPE_D, "ring alignment");
  bf16x8 b_ring[PIPE_D][4];
  auto load_group = [&](bf16x8 (&dst)[4], int64_t tile, int ks) {
    const int64_t n0 = tile << 6;
    const int64_t item_base = n0 + (lane & 15);
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      const int64_t item = item_base + f * 16;
      const __hip_bfloat16* wr = w + (size_t)min(item, V - 1) * E;
      dst[f] = *reinterpret_cast<const bf16x8*>(wr + ks * 32 + bk0);
    }
  };
  auto gidx_tile = [&](int64_t tile, int ks, int ahead) {
    // (tile, ks) advanced by `ahead` K-sections in this WG's walk
    int nks = ks + ahead;
    return tile + (int64_t)(nks / KSTEPS) * gridDim.y;
  };
  const int64_t tile0 = blockIdx.y;
  if (tile0 < n_tiles) {
#pragma unroll
    for (int d = 0; d < PIPE_D && d < KSTEPS; ++d) {
      load_group(b_ring[d], gidx_tile(tile0, 0, d), d % KSTEPS);
    }
    if constexpr (KSTEPS < PIPE_D) {
#pragma unroll
      for (int d = KSTEPS; d < PIPE_D; ++d) {
        load_group(b_ring[d], gidx_tile(tile0, 0, d), d % KSTEPS);
      }
    }
  }
  for (int64_t tile = tile0; tile < n_tiles; tile += gridDim.y) {
    f32x4 acc[MF][4];
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
#pragma unroll
      for (int f = 0; f < 4; ++f) acc[mf][f] = f32x4{0.f, 0.f, 0.f, 0.f};
    const int64_t n0 = tile << 6;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
          const int slot = ks % PIPE_D;  // compile-time after full unroll
      bf16x8 b_frag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) b_frag[f] = b_ring[slot][f];
      // prefetch the group PIPE_D sections ahead into the freed slot
      load_group(b_ring[slot], gidx_tile(tile, ks, PIPE_D), (ks + PIPE_D) % KSTEPS);
      if constexpr (!RESIDENT) {
        // LDS fragment read: lane (g = l>>4, r = l&15) reads row (mf*16+r) of
        // the WG tile at k byte (ks*64 + g*16), XOR-swizzled as written
        const int lrow_base = wave * 64 + (lane & 15);
        const int k_byte = ks * 64 + (lane >> 4) * 16;
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
          a_frag[mf][0] = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(q_lds) + lds_off(lrow_base + mf * 16, k_byte));
        }
      }
#pragma unroll
      for (int f = 0; f < 4; ++f) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
          acc[mf][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mf][RESIDENT ? ks : 0], b_frag[f], acc[mf][f], 0, 0, 0);
        }
      }
    }
    // ---- epilogue: threshold test on accumulators (common path: no hit) ----
    bool any_hit = false;
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int64_t item = n0 + f * 16 + (lane & 15);
        if (item >= V) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          any_hit |= (acc[mf][f][r] >= t_reg[mf][r]);
        }
      }
    }
    if (__builtin_amdgcn_ballot_w64(any_hit) == 0) continue;  // fast path
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int64_t item = n0 + f * 16 + (lane & 15);
        if (item >= V) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float v = acc[mf][f][r];
          if (v >= t_reg[mf][r]) {
            const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
            const int pos = atomicAdd(&counts[row], 1);
            if (pos < cap) {
              out_vals[(size_t)row * cap + pos] = v;
              out_idx[(size_t)row * cap + pos] = (int)item;
            }
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// v2 (E = 256 shape, 8-wave): the 4-wave/128-KiB-LDS variant above runs ONE
// wave per SIMD (LDS-bound occupancy), so every B-load stall parks the only
// MFMA stream on that SIMD — measured SQ_WAIT 83%, MFMA busy 9%.  This
// variant keeps the same 256-query M-tile but splits it over 8 waves
// (32 rows each): the A fragments now fit in 64 VGPRs/lane, so there is NO
// LDS at all, two waves share each SIMD (interleaved latency hiding), and
// each wave runs a depth-PIPE_D B ring.  The 8x B-tile read redundancy
// within the workgroup is served by L1/L2 (4 KB/section tile slice), and
// the 4 M-tiles' passes over W share the die-level L3 (256 MB), so HBM
// traffic stays ~one W stream.
template <int E, int PIPE_D>
__global__ __launch_bounds__(512, 2) void scored_topk_gemm_kernel_v2(
    const __hip_bfloat16* __restrict__ q,  // [M, E]
    const __hip_bfloat16* __restrict__ w,  // [V, E]
    const float* __restrict__ thresholds,  // [M]
    float* __restrict__ out_vals,          // [M, cap]
    int* __restrict__ out_idx,             // [M, cap]
    int* __restrict__ counts,              // [M]
    int M, int64_t V, int cap) {
  constexpr int KSTEPS = E / 32;
  constexpr int MF = 2;  // row-fragments per wave (wave covers 32 rows)
  static_assert(KSTEPS % PIPE_D == 0, "ring slot rotation needs KSTEPS % PIPE_D == 0");
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int m0 = blockIdx.x * 256 + wave * 32;

  // ---- resident A fragments: 32 rows x E in 8*KSTEPS VGPR quads ----
  bf16x8 a_frag[MF][KSTEPS];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int row = m0 + mf * 16 + (lane & 15);
    const __hip_bfloat16* qr = q + (size_t)min(row, M - 1) * E + (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      a_frag[mf][ks] = *reinterpret_cast<const bf16x8*>(qr + ks * 32);
    }
    if (row >= M) {
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) a_frag[mf][ks] = bf16x8{0};
    }
  }
  float t_reg[MF][4];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
      t_reg[mf][r] = (row < M) ? thresholds[row] : INFINITY;
    }
  }

  const int64_t n_tiles = (V + 63) >> 6;
  const int64_t tile0 = blockIdx.y;
  if (tile0 >= n_tiles) return;
  // last tile whose 64 items are all in range on the fast (unclamped) path
  const bool grid_tail = ((n_tiles << 6) != V);
  const int64_t tile_stride = gridDim.y;

  // per-lane W walk pointers: fragment f of section ks of tile t lives at
  //   w + (t*64 + f*16 + (lane&15))*E + ks*32 + (lane>>4)*8
  const int bk0 = (lane >> 4) * 8;
  const __hip_bfloat16* wp = w + ((size_t)(tile0 << 6) + (lane & 15)) * E + bk0;
  const size_t wp_stride = (size_t)tile_stride * 64 * E;  // elements per tile step

  bf16x8 b_ring[PIPE_D][4];
  // load group (tile, ks): 4 fragments, fast path (no clamp)
  auto load_group_fast = [&](bf16x8 (&dst)[4], const __hip_bfloat16* base, int ks) {
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      dst[f] = *reinterpret_cast<const bf16x8*>(base + (size_t)f * 16 * E + ks * 32);
    }
  };
  auto load_group_clamped = [&](bf16x8 (&dst)[4], int64_t tile, int ks) {
    const int64_t n0 = tile << 6;
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      const int64_t item = n0 + f * 16 + (lane & 15);
      const __hip_bfloat16* wr = w + (size_t)min(item, V - 1) * E;
      dst[f] = *reinterpret_cast<const bf16x8*>(wr + ks * 32 + bk0);
    }
  };
  auto load_group = [&](bf16x8 (&dst)[4], int64_t tile, const __hip_bfloat16* base, int ks) {
    if (tile >= n_tiles) return;  // ring slot past the walk: leave stale, never consumed
    if (grid_tail && tile == n_tiles - 1) {
      load_group_clamped(dst, tile, ks);
    } else {
      load_group_fast(dst, base, ks);
    }
  };

  // prologue: fill the ring PIPE_D sections ahead (may cross tile bounds)
#pragma unroll
  for (int d = 0; d < PIPE_D; ++d) {
    const int64_t t = tile0 + (int64_t)(d / KSTEPS) * tile_stride;
    load_group(b_ring[d], t, wp + (size_t)(d / KSTEPS) * wp_stride, d % KSTEPS);
  }

  for (int64_t tile = tile0; tile < n_tiles; tile += tile_stride) {
    f32x4 acc[MF][4];
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
#pragma unroll
      for (int f = 0; f < 4; ++f) acc[mf][f] = f32x4{0.f, 0.f, 0.f, 0.f};
    const int64_t n0 = tile << 6;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      const int slot = ks % PIPE_D;
      bf16x8 b_frag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) b_frag[f] = b_ring[slot][f];
      // refill the slot PIPE_D sections ahead
      {
        const int nks = ks + PIPE_D;
        const int64_t t_ahead = tile + (int64_t)(nks / KSTEPS) * tile_stride;
        load_group(b_ring[slot], t_ahead, wp + (size_t)(nks / KSTEPS) * wp_stride,
                   nks % KSTEPS);
      }
#pragma unroll
      for (int f = 0; f < 4; ++f) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
          acc[mf][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mf][ks], b_frag[f], acc[mf][f], 0, 0, 0);
        }
      }
    }
    wp += wp_stride;
    // ---- epilogue: threshold test on accumulators (common path: no hit) ----
    bool any_hit = false;
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int64_t item = n0 + f * 16 + (lane & 15);
        if (item >= V) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          any_hit |= (acc[mf][f][r] >= t_reg[mf][r]);
        }
      }
    }
    if (__builtin_amdgcn_ballot_w64(any_hit) == 0) continue;  // fast path
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int64_t item = n0 + f * 16 + (lane & 15);
        if (item >= V) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float v = acc[mf][f][r];
          if (v >= t_reg[mf][r]) {
            const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
            const int pos = atomicAdd(&counts[row], 1);
            if (pos < cap) {
              out_vals[(size_t)row * cap + pos] = v;
              out_idx[(size_t)row * cap + pos] = (int)item;
            }
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// v3 (E = 256): v2 keeps A resident but has every wave re-load the shared
// 64-item B tile (8x redundancy -> L1/L2 bound, measured 22 ms).  v3 stages
// the B tile ONCE per workgroup into LDS with cooperative
// global_load_lds (LDS-DMA, no staging VGPRs), double-buffered so the DMA
// for tile t+1 overlaps the MFMA work on tile t (guide §5 canonical GEMM).
//
// LDS image: 64 rows x 544 B (W row 512 B + 32 B junk pad).  The pad makes
// the row stride 136 dwords == 8 mod 64, so the 16 lanes of a
// ds_read_b128 group land on 16 DISTINCT banks (conflict-free); the junk
// chunks are filled from a safe dummy address (glds is lane-linear: the
// image must be written in exactly lane order, so the pad must be part of
// the stream, guide §5 glds caveat).
template <int E>
__global__ __launch_bounds__(512, 2) void scored_topk_gemm_kernel_v3(
    const __hip_bfloat16* __restrict__ q,  // [M, E]
    const __hip_bfloat16* __restrict__ w,  // [V, E]
    const float* __restrict__ thresholds,  // [M]
    float* __restrict__ out_vals,          // [M, cap]
    int* __restrict__ out_idx,             // [M, cap]
    int* __restrict__ counts,              // [M]
    int M, int64_t V, int cap) {
  constexpr int KSTEPS = E / 32;
  constexpr int MF = 2;                    // row-fragments per wave (32 rows)
  constexpr int ROW_B = E * 2 + 32;        // padded LDS row bytes (544 at E=256)
  constexpr int CHUNKS_ROW = ROW_B / 16;   // 34
  constexpr int TILE_CHUNKS = 64 * CHUNKS_ROW;  // 2176 chunks of 16 B
  constexpr int TILE_B = 64 * ROW_B;       // 34816 B per buffer
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int m0 = blockIdx.x * 256 + wave * 32;

  extern __shared__ __attribute__((aligned(16))) char smem[];  // 2 x TILE_B

  // ---- resident A fragments: 32 rows x E ----
  bf16x8 a_frag[MF][KSTEPS];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int row = m0 + mf * 16 + (lane & 15);
    const __hip_bfloat16* qr = q + (size_t)min(row, M - 1) * E + (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      a_frag[mf][ks] = *reinterpret_cast<const bf16x8*>(qr + ks * 32);
    }
    if (row >= M) {
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) a_frag[mf][ks] = bf16x8{0};
    }
  }
  float t_reg[MF][4];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
      t_reg[mf][r] = (row < M) ? thresholds[row] : INFINITY;
    }
  }

  const int64_t n_tiles = (V + 63) >> 6;
  const int64_t tile0 = blockIdx.y;
  const int64_t tile_stride = gridDim.y;
  if (tile0 >= n_tiles) return;

  // glds stage of one tile into buffer `buf`: this wave issues wave-count-
  // strided 1 KiB pieces; lane's source chunk c = piece*64 + lane.
  // item = c / 34, sub = c % 34; sub >= 32 is the pad (load chunk 0 again —
  // glds always writes its lane slot, so feed it a safe in-range address).
  auto stage_tile = [&](int buf, int64_t tile) {
    const __hip_bfloat16* wt = w + (size_t)(tile << 6) * E;
    const bool tail = ((tile << 6) + 64) > V;
    char* lds_base = smem + (size_t)buf * TILE_B;
#pragma unroll
    for (int piece = 0; piece < TILE_CHUNKS / 64 / 8; ++piece) {
      const int c = (piece * 8 + wave) * 64 + lane;
      int item = c / CHUNKS_ROW;
      int sub = c % CHUNKS_ROW;
      if (sub >= E * 2 / 16) sub = 0;  // pad chunk: safe dummy source
      if (tail) {
        const int64_t gitem = (tile << 6) + item;
        item -= (int)(gitem >= V ? (gitem - (V - 1)) : 0);  // clamp into range
      }
      const char* src = reinterpret_cast<const char*>(wt) + (size_t)item * (E * 2) + sub * 16;
      __builtin_amdgcn_global_load_lds(
          (const void*)src, (void*)(lds_base + (size_t)(piece * 8 + wave) * 1024), 16, 0, 0);
    }
  };
  // TILE_CHUNKS/64 = 34 wave-pieces per tile; 8 waves round-robin -> waves
  // 0,1 carry 5 pieces, the rest 4 (34 = 4*8 + 2)
  auto stage_rem = [&](int buf, int64_t tile) {
    const int base_piece = (TILE_CHUNKS / 64 / 8) * 8;  // 32
    if (wave < (TILE_CHUNKS / 64) - base_piece) {
      const __hip_bfloat16* wt = w + (size_t)(tile << 6) * E;
      const bool tail = ((tile << 6) + 64) > V;
      char* lds_base = smem + (size_t)buf * TILE_B;
      const int c = (base_piece + wave) * 64 + lane;
      int item = c / CHUNKS_ROW;
      int sub = c % CHUNKS_ROW;
      if (sub >= E * 2 / 16) sub = 0;
      if (tail) {
        const int64_t gitem = (tile << 6) + item;
        item -= (int)(gitem >= V ? (gitem - (V - 1)) : 0);
      }
      const char* src = reinterpret_cast<const char*>(wt) + (size_t)item * (E * 2) + sub * 16;
      __builtin_amdgcn_global_load_lds(
          (const void*)src, (void*)(lds_base + (size_t)(base_piece + wave) * 1024), 16, 0, 0);
    }
  };

  stage_tile(0, tile0);
  stage_rem(0, tile0);
  __syncthreads();  // drains the glds (vmcnt(0) folded into the barrier)

  int cur = 0;
  for (int64_t tile = tile0; tile < n_tiles; tile += tile_stride) {
    const int64_t nxt = tile + tile_stride;
    if (nxt < n_tiles) {
      stage_tile(cur ^ 1, nxt);
      stage_rem(cur ^ 1, nxt);
    }
    const char* bbuf = smem + (size_t)cur * TILE_B;
    f32x4 acc[MF][4];
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
#pragma unroll
      for (int f = 0; f < 4; ++f) acc[mf][f] = f32x4{0.f, 0.f, 0.f, 0.f};
    const int64_t n0 = tile << 6;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      bf16x8 b_frag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int item = f * 16 + (lane & 15);
        b_frag[f] = *reinterpret_cast<const bf16x8*>(
            bbuf + (size_t)item * ROW_B + ks * 64 + (lane >> 4) * 16);
      }
#pragma unroll
      for (int f = 0; f < 4; ++f) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
          acc[mf][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mf][ks], b_frag[f], acc[mf][f], 0, 0, 0);
        }
      }
    }
    // ---- epilogue: threshold test (common path: no hit) ----
    bool any_hit = false;
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int64_t item = n0 + f * 16 + (lane & 15);
        if (item >= V) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          any_hit |= (acc[mf][f][r] >= t_reg[mf][r]);
        }
      }
    }
    if (__builtin_amdgcn_ballot_w64(any_hit) != 0) {
#pragma unroll
      for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
        for (int f = 0; f < 4; ++f) {
          const int64_t item = n0 + f * 16 + (lane & 15);
          if (item >= V) continue;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float v = acc[mf][f][r];
            if (v >= t_reg[mf][r]) {
              const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
              const int pos = atomicAdd(&counts[row], 1);
              if (pos < cap) {
                out_vals[(size_t)row * cap + pos] = v;
                out_idx[(size_t)row * cap + pos] = (int)item;
              }
            }
          }
        }
      }
    }
    __syncthreads();  // buffer swap guard (also drains next tile's glds)
    cur ^= 1;
  }
}

// ---------------------------------------------------------------------------
// v4 (E = 256): v3 with the barrier drain removed.  v3's __syncthreads
// folds a vmcnt(0) whenever a glds is in flight, so every tile waits for
// the NEXT tile's full 34 KB DMA before any wave proceeds (guide §5: the
// structural ~20% stall of the 2-buffer + __syncthreads form).  v4 keeps
// THREE LDS buffers with 2 tiles of DMA in flight, raw s_barrier, and a
// per-wave counted s_waitcnt vmcnt(N) right before the fragment reads —
// the canonical glds pipeline (guide: +83% over serial vs +40%).
//
// Protocol per iteration i (cur = i % 3):
//   counted vmcnt(own pieces of 1 newer tile)  -> own glds for buf[cur] done
//                                                 (epilogue stores are OLDER
//                                                 than the newest stage, so
//                                                 the counted wait drains
//                                                 them for free)
//   raw s_barrier                              -> everyone's glds for buf[cur]
//                                                 done AND everyone finished
//                                                 reading buf[(i+2)%3] in i-1
//   compute buf[cur] + epilogue
//   stage tile i+2 into buf[(i+2)%3] (LAST, so it is the newest VMEM)
template <int N>
__device__ __forceinline__ void waitcnt_vm() {
  if constexpr (N <= 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 1) asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
  else if constexpr (N == 2) asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  else if constexpr (N == 3) asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else if constexpr (N == 5) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
  else if constexpr (N == 6) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else if constexpr (N == 8) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else if constexpr (N == 10) asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
  else if constexpr (N == 17) asm volatile("s_waitcnt vmcnt(17)" ::: "memory");
  else static_assert(N == 0, "add a vmcnt immediate case");
}

// ABLATE: 0 full; 1 no-stage; 2 no-mfma; 3 no-epilogue; 4 no-epilogue +
// no-barrier/vmcnt (pure ds_read+MFMA loop); 5 no-epilogue + no-stage
template <int E, int MF, int NBUF = 3, int ABLATE = 0, int LOADERS = 0, int XCD_MAP = 1,
          int SPREAD = 1, int LDSEPI = 1>
__global__ __launch_bounds__(512, 2) void scored_topk_gemm_kernel_v4(
    const __hip_bfloat16* __restrict__ q,  // [M, E]
    const __hip_bfloat16* __restrict__ w,  // [V, E]
    const float* __restrict__ thresholds,  // [M]
    float* __restrict__ out_vals,          // [M, cap]
    int* __restrict__ out_idx,             // [M, cap]
    int* __restrict__ counts,              // [M]
    int M, int64_t V64, int cap) {
  constexpr int KSTEPS = E / 32;
  constexpr int ROW_B = E * 2 + 32;             // 544
  constexpr int CHUNKS_ROW = ROW_B / 16;        // 34
  constexpr int TILE_PIECES = 64 * CHUNKS_ROW / 64;  // 34 wave-pieces
  constexpr int TILE_B = 64 * ROW_B;
  constexpr size_t TILE_W = (size_t)64 * E * 2;  // W bytes per tile (unpadded)
  // LDS hit-append epilogue (LDSEPI): per-wave per-row candidate lists.
  // Hits are appended with a ds_add_rtn + ds_write (~100 cyc, no vmem)
  // instead of a global atomicAdd whose result feeds the store (~1 L2
  // round trip serialized under the per-tile barrier — PMC showed 53% of
  // wave cycles parked with the global path).  The lists flush to the
  // global candidate buffer ONCE per block walk, one lane-parallel
  // atomicAdd per row.  RCAP=8 covers the expected ~1 hit/row/walk at
  // serving thresholds; overflow falls back to the direct global append
  // (correct, just slower — only matters when few stripes make walks
  // long, e.g. very large M).
  constexpr int RCAP = 8;
  constexpr int ROWS_W = 16 * MF;  // rows owned by one wave
  constexpr size_t EPI_PER_WAVE = (size_t)ROWS_W * 4 + (size_t)ROWS_W * RCAP * 8;
  const int V = (int)V64;  // host asserts V < 2^31 (out_idx is int32 anyway);
                           // 32-bit index math saves ~30 VGPRs at MF=4
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  // XCD-aware (m_tile, stripe) assignment.  Workgroups dispatch round-robin
  // over the 8 XCDs, so the naive (x = m_tile, y = stripe) grid puts the
  // m_tiles that share a B stripe on DIFFERENT XCDs: every XCD streams the
  // whole item table through its own (private) L2 and the kernel is
  // HBM-bound on the 8x re-read (measured 5.6 ms at B=1024/V=10M — the
  // 40 GB the 8 XCDs pull at ~7 TB/s).  Remap (bijective, guide §"XCD
  // swizzle") so each XCD owns a contiguous band of stripes together with
  // ALL their m_tiles: each B tile is pulled from HBM by exactly one XCD
  // and the co-resident m_tile blocks hit it in that XCD's L2.
  int bx = blockIdx.x, by = blockIdx.y;
  if constexpr (XCD_MAP) {
    const int nwg = (int)(gridDim.x * gridDim.y);
    const int orig = (int)(blockIdx.y * gridDim.x + blockIdx.x);
    const int qq = nwg >> 3, rr = nwg & 7, xcd = orig & 7, slot = orig >> 3;
    const int rid =
        (xcd < rr ? xcd * (qq + 1) : rr * (qq + 1) + (xcd - rr) * qq) + slot;
    bx = rid % (int)gridDim.x;  // m_tile: fastest within an XCD's band
    by = rid / (int)gridDim.x;  // stripe
  }
  const int m0 = bx * (128 * MF) + wave * (16 * MF);

  extern __shared__ __attribute__((aligned(16))) char smem[];  // 3 x TILE_B

  bf16x8 a_frag[MF][KSTEPS];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int row = m0 + mf * 16 + (lane & 15);
    const __hip_bfloat16* qr = q + (size_t)min(row, M - 1) * E + (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      a_frag[mf][ks] = *reinterpret_cast<const bf16x8*>(qr + ks * 32);
    }
    if (row >= M) {
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) a_frag[mf][ks] = bf16x8{0};
    }
  }
  // thresholds packed two-per-register as bf16, rounded toward -inf: the
  // packed value never exceeds the fp32 threshold, so the (5x oversized)
  // candidate capacity absorbs the few extra admits and the exact final
  // top-k is unchanged.  Saves MF*2 VGPRs — the margin that keeps the
  // E=256 spread-staging variant spill-free.
  unsigned tpk[MF][2];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
      const float tf = (row < M) ? thresholds[row] : INFINITY;
      const unsigned u = __float_as_uint(tf);
      unsigned short h = (unsigned short)(u >> 16);
      if ((u & 0xFFFFu) && (u & 0x80000000u)) ++h;  // negative: round away from 0
      if ((r & 1) == 0) tpk[mf][r >> 1] = h;
      else tpk[mf][r >> 1] |= (unsigned)h << 16;
    }
  }
  const auto t_reg_at = [&](int mf, int r) -> float {
    return __uint_as_float(((tpk[mf][r >> 1] >> ((r & 1) * 16)) & 0xFFFFu) << 16);
  };
  // drain the A/threshold loads so they never mix into the glds counting
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  const int n_tiles = (V + 63) >> 6;
  const int tile0 = by;
  const int tile_stride = gridDim.y;
  if (tile0 >= n_tiles) return;

  auto stage_piece = [&](int buf, int tile, bool tail, int piece) {
    const char* wt = reinterpret_cast<const char*>(w) + (size_t)((unsigned)tile << 6) * (E * 2);
    char* lds_base = smem + (size_t)buf * TILE_B;
    const int c = piece * 64 + lane;
    int item = c / CHUNKS_ROW;
    int sub = c % CHUNKS_ROW;
    if (sub >= E * 2 / 16) sub = 0;
    if (tail) {
      const int gitem = (tile << 6) + item;
      item -= (gitem >= V ? (gitem - (V - 1)) : 0);
    }
    const unsigned off = (unsigned)item * (E * 2) + sub * 16;
    __builtin_amdgcn_global_load_lds(
        (const void*)(wt + off), (void*)(lds_base + (size_t)piece * 1024), 16, 0, 0);
  };
  auto stage_tile = [&](int buf, int tile) {
    constexpr int NLOAD = (LOADERS == 0) ? 8 : LOADERS;
    if (LOADERS != 0 && wave >= LOADERS) return;
    const bool tail = ((tile << 6) + 64) > V;
    for (int piece = wave; piece < TILE_PIECES; piece += NLOAD) {
      stage_piece(buf, tile, tail, piece);
    }
  };
  // spread-staging (SPREAD=1, guide T3/T4): tile+2's glds are issued one
  // piece per wave per f-column instead of one burst after the tile's
  // compute, so the VMEM pipe is fed continuously under the MFMA runs.
  // The counted-vmcnt accounting at the tile-top wait is unchanged: by the
  // time a wave reaches the wait for tile t+1, tile t+2 is fully issued
  // either way.  The per-lane within-tile W offsets are tile-invariant for
  // a fixed (wave, lane), so they are precomputed HERE: inside the f-loop
  // a glds issue is just uniform-scalar base + 32-bit voffset, keeping the
  // address math out of the MFMA-region live set (64-bit per-lane address
  // chains there spill at E=256).
  constexpr bool DO_SPREAD = (SPREAD == 1) && (LOADERS == 0) && NBUF >= 3;
  // five per-lane offsets, each < 64*E*2 <= 32 KiB, packed as u16 pairs
  unsigned spread_pk[DO_SPREAD ? 3 : 1] = {0};
  if constexpr (DO_SPREAD) {
#pragma unroll
    for (int f = 0; f < 5; ++f) {
      const int piece = wave + f * 8;
      const int c = piece * 64 + lane;
      int item = c / CHUNKS_ROW;
      int sub = c % CHUNKS_ROW;
      if (sub >= E * 2 / 16) sub = 0;
      const unsigned off = (unsigned)item * (E * 2) + sub * 16;
      spread_pk[f >> 1] |= off << ((f & 1) * 16);
    }
  }
  const auto spread_off = [&](int f) -> unsigned {
    return (spread_pk[f >> 1] >> ((f & 1) * 16)) & 0xFFFFu;
  };

  unsigned* ecnt = nullptr;
  float* ebufv = nullptr;
  int* ebufi = nullptr;
  if constexpr (LDSEPI) {
    char* epi = smem + (size_t)NBUF * TILE_B + (size_t)wave * EPI_PER_WAVE;
    ecnt = reinterpret_cast<unsigned*>(epi);
    ebufv = reinterpret_cast<float*>(epi + ROWS_W * 4);
    ebufi = reinterpret_cast<int*>(epi + ROWS_W * 4 + ROWS_W * RCAP * 4);
    for (int lr = lane; lr < ROWS_W; lr += WAVE) ecnt[lr] = 0;
  }

#pragma unroll
  for (int d = 0; d < NBUF - 1; ++d) {
    if (tile0 + d * tile_stride < n_tiles) stage_tile(d, tile0 + d * tile_stride);
  }

  int cur = 0;
  for (int tile = tile0; tile < n_tiles; tile += tile_stride) {
    // own glds for buf[cur] complete; allow NBUF-2 newer tiles in flight
    // (when they were actually staged — at the walk's tail, full drain)
    if constexpr (ABLATE != 4) {
      constexpr int NLOAD = (LOADERS == 0) ? 8 : LOADERS;
      constexpr int P_HI = (TILE_PIECES + NLOAD - 1) / NLOAD;
      constexpr int P_LO = TILE_PIECES / NLOAD;
      const bool is_loader = (LOADERS == 0) || (wave < LOADERS);
      if (is_loader) {
        if (tile + (int)(NBUF - 2) * tile_stride < n_tiles) {
          if (wave < (TILE_PIECES % NLOAD)) {
            waitcnt_vm<P_HI * (NBUF - 2)>();
          } else {
            waitcnt_vm<P_LO * (NBUF - 2)>();
          }
        } else {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
      }
      __builtin_amdgcn_s_barrier();
    }

    const char* bbuf = smem + (size_t)cur * TILE_B;
    const int n0 = tile << 6;
    const int spread_t2 = tile + (NBUF - 1) * tile_stride;
    const int spread_buf = (cur + NBUF - 1) % NBUF;
    // one 16-item column fragment at a time: acc live set = MF quads (not
    // MF x 4), which is what lets MF = 4 (512-row M-tile) fit in 256 VGPRs.
    // B fragments are prefetched a full column ahead (ping-pong bfr[2][8])
    // so the 32-MFMA run per column never parks on lgkmcnt — without this
    // the ds_read -> 4xMFMA chains serialize (measured 5x off the MFMA
    // floor with DMA fully hidden).
    bf16x8 bfr[2][KSTEPS];
    {
      const char* bcol0 = bbuf + (size_t)(lane & 15) * ROW_B + (lane >> 4) * 16;
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        bfr[0][ks] = *reinterpret_cast<const bf16x8*>(bcol0 + ks * 64);
      }
    }
#pragma unroll
    for (int f = 0; f < 4 && ABLATE != 2; ++f) {
      if (f < 3) {
        const char* bcoln =
            bbuf + (size_t)((f + 1) * 16 + (lane & 15)) * ROW_B + (lane >> 4) * 16;
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks) {
          bfr[(f + 1) & 1][ks] = *reinterpret_cast<const bf16x8*>(bcoln + ks * 64);
        }
      }
      if constexpr (DO_SPREAD) {
        if (spread_t2 < n_tiles && ABLATE != 1 && ABLATE != 5) {
          // tail tiles: clamp the UNIFORM tile base so wt2+off stays in
          // bounds (the duplicated rows staged for OOB items are masked by
          // the item < V epilogue guard) — no per-lane tail math in the
          // MFMA-region live set
          const size_t woff =
              min((size_t)((unsigned)spread_t2 << 6) * (E * 2), (size_t)V * (E * 2) - TILE_W);
          const char* wt2 = reinterpret_cast<const char*>(w) + woff;
          char* sl = smem + (size_t)spread_buf * TILE_B;
          const int piece = wave + f * 8;
          if (piece < TILE_PIECES) {
            __builtin_amdgcn_global_load_lds((const void*)(wt2 + spread_off(f)),
                                             (void*)(sl + (size_t)piece * 1024), 16, 0, 0);
          }
          if (f == 3 && wave + 32 < TILE_PIECES) {
            __builtin_amdgcn_global_load_lds((const void*)(wt2 + spread_off(4)),
                                             (void*)(sl + (size_t)(wave + 32) * 1024), 16, 0,
                                             0);
          }
        }
      }
      f32x4 acc[MF];
#pragma unroll
      for (int mf = 0; mf < MF; ++mf) acc[mf] = f32x4{0.f, 0.f, 0.f, 0.f};
      const int item = n0 + f * 16 + (lane & 15);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
          acc[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mf][ks], bfr[f & 1][ks], acc[mf], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      bool any_hit = false;
      if (item < V) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
          for (int r = 0; r < 4; ++r) any_hit |= (acc[mf][r] >= t_reg_at(mf, r));
        }
      }
      if constexpr (ABLATE == 3 || ABLATE == 4 || ABLATE == 5) {
        // keep acc live without the atomic epilogue (never true at runtime)
        if (any_hit && cap == -1) counts[0] = 1;
        continue;
      }
      if (__builtin_amdgcn_ballot_w64(any_hit) != 0) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float v = acc[mf][r];
            if (item < V && v >= t_reg_at(mf, r)) {
              const int lr = mf * 16 + (lane >> 4) * 4 + r;
              if constexpr (LDSEPI) {
                const unsigned n = atomicAdd(&ecnt[lr], 1u);  // ds_add_rtn
                if (n < (unsigned)RCAP) {
                  ebufv[lr * RCAP + (int)n] = v;
                  ebufi[lr * RCAP + (int)n] = item;
                } else {
                  // list full: push counts past capacity so the host's
                  // exact-rescore fallback owns this row (result unused —
                  // a fire-and-forget atomic, no round trip in the loop).
                  // The host only picks this kernel when overflow is rare.
                  atomicAdd(&counts[m0 + lr], cap + 1);
                }
              } else {
                const int row = m0 + lr;
                const int pos = atomicAdd(&counts[row], 1);
                if (pos < cap) {
                  // 32-bit offset (host checks M*cap*4 < 2^31): keeps the
                  // store at SGPR-base + voffset so the output pointers
                  // never occupy VGPR pairs across the MFMA region
                  const int off = row * cap + pos;
                  out_vals[off] = v;
                  out_idx[off] = item;
                }
              }
            }
          }
        }
      }
    }
    // without spread-staging: stage tile+2 in one burst LAST so the next
    // iteration's counted vmcnt drains this iteration's (older) epilogue
    // stores together with tile+1's glds
    if constexpr (!DO_SPREAD && ABLATE != 1 && ABLATE != 5) {
      const int t2 = tile + (NBUF - 1) * tile_stride;
      if (t2 < n_tiles) stage_tile((cur + NBUF - 1) % NBUF, t2);
    }
    cur = (cur + 1) % NBUF;
  }
  if constexpr (LDSEPI && ABLATE != 3 && ABLATE != 4 && ABLATE != 5) {
    // walk-end flush: lane-parallel, one global atomicAdd per owned row
    for (int lr = lane; lr < ROWS_W; lr += WAVE) {
      const unsigned c = ecnt[lr];
      const int m_ = (int)min(c, (unsigned)RCAP);
      const int row = m0 + lr;
      if (m_ > 0 && row < M) {
        const int base = atomicAdd(&counts[row], m_);
        for (int j = 0; j < m_; ++j) {
          const int pos = base + j;
          if (pos < cap) {
            const int off = row * cap + pos;
            out_vals[off] = ebufv[lr * RCAP + j];
            out_idx[off] = ebufi[lr * RCAP + j];
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// fp8 (OCP e4m3) variant of the v4 pipeline for the config-5 serve path:
// both operands quantized e4m3 (global amax scales folded into the
// thresholds by the host; the raw fp32 accumulators come back and the host
// multiplies by scale AFTER selection — monotonic, so selection is
// unchanged).  Same glds triple-buffer + raw-barrier + counted-vmcnt
// structure; W rows are E bytes (half the bf16 stream), LDS rows padded to
// E+48 so stride/4 == 12 (mod 64) keeps the two 32-lane ds_read_b64 groups
// conflict-free.  MFMA: v_mfma_f32_16x16x32_fp8_fp8 (i64 operands = 8
// packed e4m3; same fragment index map as the bf16 form, bf16-rate).
template <int E, int MF>
__global__ __launch_bounds__(512, 2) void scored_topk_gemm_fp8_kernel(
    const unsigned char* __restrict__ q,  // [M, E] e4m3
    const unsigned char* __restrict__ w,  // [V, E] e4m3
    const float* __restrict__ thresholds,  // [M] (pre-divided by scale)
    float* __restrict__ out_vals,          // [M, cap] RAW accumulators
    int* __restrict__ out_idx,             // [M, cap]
    int* __restrict__ counts,              // [M]
    int M, int64_t V64, int cap) {
  constexpr int KSTEPS = E / 32;
  constexpr int ROW_B = E + 48;                 // 304 at E=256
  constexpr int CHUNKS_ROW = ROW_B / 16;        // 19
  constexpr int TILE_PIECES = 64 * CHUNKS_ROW / 64;
  constexpr int TILE_B = 64 * ROW_B;
  const int V = (int)V64;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  // XCD-aware (m_tile, stripe) assignment.  Workgroups dispatch round-robin
  // over the 8 XCDs, so the naive (x = m_tile, y = stripe) grid puts the
  // m_tiles that share a B stripe on DIFFERENT XCDs: every XCD streams the
  // whole item table through its own (private) L2 and the kernel is
  // HBM-bound on the 8x re-read (measured 5.6 ms at B=1024/V=10M — the
  // 40 GB the 8 XCDs pull at ~7 TB/s).  Remap (bijective, guide §"XCD
  // swizzle") so each XCD owns a contiguous band of stripes together with
  // ALL their m_tiles: each B tile is pulled from HBM by exactly one XCD
  // and the co-resident m_tile blocks hit it in that XCD's L2.
  int bx = blockIdx.x, by = blockIdx.y;
  {  // always on for the fp8 kernel
    const int nwg = (int)(gridDim.x * gridDim.y);
    const int orig = (int)(blockIdx.y * gridDim.x + blockIdx.x);
    const int qq = nwg >> 3, rr = nwg & 7, xcd = orig & 7, slot = orig >> 3;
    const int rid =
        (xcd < rr ? xcd * (qq + 1) : rr * (qq + 1) + (xcd - rr) * qq) + slot;
    bx = rid % (int)gridDim.x;  // m_tile: fastest within an XCD's band
    by = rid / (int)gridDim.x;  // stripe
  }
  const int m0 = bx * (128 * MF) + wave * (16 * MF);

  extern __shared__ __attribute__((aligned(16))) char smem[];  // 3 x TILE_B

  long a_frag[MF][KSTEPS];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int row = m0 + mf * 16 + (lane & 15);
    const unsigned char* qr = q + (size_t)min(row, M - 1) * E + (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      a_frag[mf][ks] = *reinterpret_cast<const long*>(qr + ks * 32);
    }
    if (row >= M) {
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) a_frag[mf][ks] = 0;
    }
  }
  float t_reg[MF][4];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
      t_reg[mf][r] = (row < M) ? thresholds[row] : INFINITY;
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  const int n_tiles = (V + 63) >> 6;
  const int tile0 = by;
  const int tile_stride = gridDim.y;
  if (tile0 >= n_tiles) return;

  auto stage_tile = [&](int buf, int tile) {
    const char* wt = reinterpret_cast<const char*>(w) + (size_t)((unsigned)tile << 6) * E;
    const bool tail = ((tile << 6) + 64) > V;
    char* lds_base = smem + (size_t)buf * TILE_B;
    for (int piece = wave; piece < TILE_PIECES; piece += 8) {
      const int c = piece * 64 + lane;
      int item = c / CHUNKS_ROW;
      int sub = c % CHUNKS_ROW;
      if (sub >= E / 16) sub = 0;
      if (tail) {
        const int gitem = (tile << 6) + item;
        item -= (gitem >= V ? (gitem - (V - 1)) : 0);
      }
      const unsigned off = (unsigned)item * E + sub * 16;
      __builtin_amdgcn_global_load_lds(
          (const void*)(wt + off), (void*)(lds_base + (size_t)piece * 1024), 16, 0, 0);
    }
  };

  stage_tile(0, tile0);
  if (tile0 + tile_stride < n_tiles) stage_tile(1, tile0 + tile_stride);

  int cur = 0;
  for (int tile = tile0; tile < n_tiles; tile += tile_stride) {
    constexpr int P_HI = (TILE_PIECES + 7) / 8;
    constexpr int P_LO = TILE_PIECES / 8;
    if (tile + tile_stride < n_tiles) {
      if (wave < (TILE_PIECES & 7)) {
        waitcnt_vm<P_HI>();
      } else {
        waitcnt_vm<P_LO>();
      }
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    const char* bbuf = smem + (size_t)cur * TILE_B;
    const int n0 = tile << 6;
    long bfr[2][KSTEPS];
    {
      const char* bcol0 = bbuf + (size_t)(lane & 15) * ROW_B + (lane >> 4) * 8;
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        bfr[0][ks] = *reinterpret_cast<const long*>(bcol0 + ks * 32);
      }
    }
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      if (f < 3) {
        const char* bcoln = bbuf + (size_t)((f + 1) * 16 + (lane & 15)) * ROW_B + (lane >> 4) * 8;
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks) {
          bfr[(f + 1) & 1][ks] = *reinterpret_cast<const long*>(bcoln + ks * 32);
        }
      }
      f32x4 acc[MF];
#pragma unroll
      for (int mf = 0; mf < MF; ++mf) acc[mf] = f32x4{0.f, 0.f, 0.f, 0.f};
      const int item = n0 + f * 16 + (lane & 15);
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
          acc[mf] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              a_frag[mf][ks], bfr[f & 1][ks], acc[mf], 0, 0, 0);
        }
      }
      bool any_hit = false;
      if (item < V) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
          for (int r = 0; r < 4; ++r) any_hit |= (acc[mf][r] >= t_reg[mf][r]);
        }
      }
      if (__builtin_amdgcn_ballot_w64(any_hit) != 0) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float v = acc[mf][r];
            if (item < V && v >= t_reg[mf][r]) {
              const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
              const int pos = atomicAdd(&counts[row], 1);
              if (pos < cap) {
                out_vals[(size_t)row * cap + pos] = v;
                out_idx[(size_t)row * cap + pos] = item;
              }
            }
          }
        }
      }
    }
    {
      const int t2 = tile + 2 * tile_stride;
      if (t2 < n_tiles) stage_tile((cur + 2) % 3, t2);
    }
    cur = (cur + 1) % 3;
  }
}

// ---------------------------------------------------------------------------
// v6 (E = 256): occupancy experiment — MF=2 (116 VGPRs) + 2 LDS buffers
// (70 KB) lets TWO workgroups co-reside per CU (4 waves/SIMD), so one WG's
// barrier/atomic stalls overlap the other's MFMA stream.  Costs 2x the W
// passes of MF=4 (4 M-tiles), bets on L3 dedup of the co-walking tiles.
template <int E>
__global__ __launch_bounds__(512, 4) void scored_topk_gemm_kernel_v6(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ w,
    const float* __restrict__ thresholds, float* __restrict__ out_vals,
    int* __restrict__ out_idx, int* __restrict__ counts, int M, int64_t V64, int cap) {
  constexpr int LOADERS = 0;  // all waves stage (the shared stage_tile text)
  constexpr int MF = 2;
  constexpr int KSTEPS = E / 32;
  constexpr int ROW_B = E * 2 + 32;
  constexpr int CHUNKS_ROW = ROW_B / 16;
  constexpr int TILE_PIECES = 64 * CHUNKS_ROW / 64;
  constexpr int TILE_B = 64 * ROW_B;
  const int V = (int)V64;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int m0 = blockIdx.x * (128 * MF) + wave * (16 * MF);

  extern __shared__ __attribute__((aligned(16))) char smem[];  // 2 x TILE_B

  bf16x8 a_frag[MF][KSTEPS];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int row = m0 + mf * 16 + (lane & 15);
    const __hip_bfloat16* qr = q + (size_t)min(row, M - 1) * E + (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) a_frag[mf][ks] = *reinterpret_cast<const bf16x8*>(qr + ks * 32);
    if (row >= M) {
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) a_frag[mf][ks] = bf16x8{0};
    }
  }
  float t_reg[MF][4];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
      t_reg[mf][r] = (row < M) ? thresholds[row] : INFINITY;
    }

  const int n_tiles = (V + 63) >> 6;
  const int tile0 = blockIdx.y;
  const int tile_stride = gridDim.y;
  if (tile0 >= n_tiles) return;

  auto stage_piece = [&](int buf, int tile, bool tail, int piece) {
    const char* wt = reinterpret_cast<const char*>(w) + (size_t)((unsigned)tile << 6) * (E * 2);
    char* lds_base = smem + (size_t)buf * TILE_B;
    const int c = piece * 64 + lane;
    int item = c / CHUNKS_ROW;
    int sub = c % CHUNKS_ROW;
    if (sub >= E * 2 / 16) sub = 0;
    if (tail) {
      const int gitem = (tile << 6) + item;
      item -= (gitem >= V ? (gitem - (V - 1)) : 0);
    }
    const unsigned off = (unsigned)item * (E * 2) + sub * 16;
    __builtin_amdgcn_global_load_lds(
        (const void*)(wt + off), (void*)(lds_base + (size_t)piece * 1024), 16, 0, 0);
  };
  auto stage_tile = [&](int buf, int tile) {
    constexpr int NLOAD = (LOADERS == 0) ? 8 : LOADERS;
    if (LOADERS != 0 && wave >= LOADERS) return;
    const bool tail = ((tile << 6) + 64) > V;
    for (int piece = wave; piece < TILE_PIECES; piece += NLOAD) {
      stage_piece(buf, tile, tail, piece);
    }
  };

  stage_tile(0, tile0);
  int cur = 0;
  for (int tile = tile0; tile < n_tiles; tile += tile_stride) {
    const int nxt = tile + tile_stride;
    if (nxt < n_tiles) stage_tile(cur ^ 1, nxt);
    __syncthreads();  // drains buf[cur]'s glds (vmcnt folded by hipcc)
    const char* bbuf = smem + (size_t)cur * TILE_B;
    const int n0 = tile << 6;
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      // single-buffer fragment reads: at 4 waves/SIMD the ds_read latency
      // hides behind the other waves (TLP), keeping VGPRs at the
      // occupancy-4 cap
      const char* bcol = bbuf + (size_t)(f * 16 + (lane & 15)) * ROW_B + (lane >> 4) * 16;
      bf16x8 bfr[KSTEPS];
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) bfr[ks] = *reinterpret_cast<const bf16x8*>(bcol + ks * 64);
      f32x4 acc[MF];
#pragma unroll
      for (int mf = 0; mf < MF; ++mf) acc[mf] = f32x4{0.f, 0.f, 0.f, 0.f};
      const int item = n0 + f * 16 + (lane & 15);
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks)
#pragma unroll
        for (int mf = 0; mf < MF; ++mf)
          acc[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[mf][ks], bfr[ks], acc[mf], 0, 0, 0);
      bool any_hit = false;
      if (item < V) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf)
#pragma unroll
          for (int r = 0; r < 4; ++r) any_hit |= (acc[mf][r] >= t_reg[mf][r]);
      }
      if (__builtin_amdgcn_ballot_w64(any_hit) != 0) {
#pragma unroll
        for (int mf = 0; mf < MF; ++mf)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float v = acc[mf][r];
            if (item < V && v >= t_reg[mf][r]) {
              const int row = m0 + mf * 16 + (lane >> 4) * 4 + r;
              const int pos = atomicAdd(&counts[row], 1);
              if (pos < cap) {
                out_vals[(size_t)row * cap + pos] = v;
                out_idx[(size_t)row * cap + pos] = item;
              }
            }
          }
      }
    }
    __syncthreads();  // buffer-swap guard
    cur ^= 1;
  }
}

}  // namespace

std::vector<torch::Tensor> scored_topk_gemm_fp8(torch::Tensor q, torch::Tensor w,
                                                torch::Tensor thresholds, int64_t capacity) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 2 && q.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.dim() == 2 && w.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kFloat8_e4m3fn && w.scalar_type() == torch::kFloat8_e4m3fn,
              "scored_topk_gemm_fp8 wants e4m3 operands");
  const int M = (int)q.size(0);
  const int E = (int)q.size(1);
  const int64_t V = w.size(0);
  TORCH_CHECK(w.size(1) == E, "dim mismatch");
  TORCH_CHECK(V < (int64_t)INT32_MAX, "catalog must fit int32 indices");
  TORCH_CHECK(E == 256 || E == 128, "scored_topk_gemm_fp8 supports E in {128, 256}");
  auto opts_f = q.options().dtype(torch::kFloat32);
  auto opts_i = q.options().dtype(torch::kInt32);
  auto out_vals = torch::full({M, capacity}, -std::numeric_limits<float>::infinity(), opts_f);
  auto out_idx = torch::zeros({M, capacity}, opts_i);
  auto counts = torch::zeros({M}, opts_i);
  auto thr = thresholds.to(torch::kFloat32).contiguous();
  const int m_tile_rows = 512;
  const int m_tiles = (M + m_tile_rows - 1) / m_tile_rows;
  int stripes = (int)std::min<int64_t>((V + 63) / 64, std::max(1, 4096 / m_tiles));
  dim3 grid(m_tiles, stripes);
  auto stream = at::cuda::getCurrentHIPStream();
#define LAUNCH_FP8(EE)                                                                       hipLaunchKernelGGL((scored_topk_gemm_fp8_kernel<EE, 4>), grid, dim3(512),                                     3 * 64 * (EE + 48), stream,                                                                reinterpret_cast<const unsigned char*>(q.data_ptr()),                                      reinterpret_cast<const unsigned char*>(w.data_ptr()),                                      thr.data_ptr<float>(), out_vals.data_ptr<float>(),                                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,                                     (int)capacity)
  if (E == 256) {
    LAUNCH_FP8(256);
  } else {
    LAUNCH_FP8(128);
  }
#undef LAUNCH_FP8
  return {out_vals, out_idx, counts};
}

std::vector<torch::Tensor> scored_topk_gemm(torch::Tensor q, torch::Tensor w,
                                            torch::Tensor thresholds, int64_t capacity) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 2 && q.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.dim() == 2 && w.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16 && w.scalar_type() == torch::kBFloat16);
  const int M = (int)q.size(0);
  const int E = (int)q.size(1);
  const int64_t V = w.size(0);
  TORCH_CHECK(w.size(1) == E, "dim mismatch");
  TORCH_CHECK(V < (int64_t)INT32_MAX, "catalog must fit int32 indices");
  auto opts_f = q.options().dtype(torch::kFloat32);
  auto opts_i = q.options().dtype(torch::kInt32);
  auto out_vals = torch::full({M, capacity}, -std::numeric_limits<float>::infinity(), opts_f);
  auto out_idx = torch::zeros({M, capacity}, opts_i);
  auto counts = torch::zeros({M}, opts_i);
  auto thr = thresholds.to(torch::kFloat32).contiguous();
  TORCH_CHECK((int64_t)M * capacity * 4 < (int64_t)INT32_MAX,
              "M*capacity too large for 32-bit candidate offsets");
  static const char* variant = std::getenv("REPLAY_AMD_STG_VARIANT");
  if (variant != nullptr && variant[0] == '\0') variant = nullptr;  // empty = default
  const bool legacy = (variant != nullptr && variant[0] >= '1' && variant[0] < '4');
  // v4 defaults: 512-row M-tile at E=256 (MF=4), 1024-row at E<=128 (MF=8);
  // wider M-tiles divide the number of passes over the streamed item table
  const bool v6 = (variant != nullptr && variant[0] == 'o');
  const int v4_mf = ((variant != nullptr && variant[0] == '5') || v6) ? 2 : (E == 64 ? 8 : 4);
  const int m_tile_rows = legacy ? 256 : 128 * v4_mf;
  const int m_tiles = (M + m_tile_rows - 1) / m_tile_rows;
  // fill 256 CUs x ~4 blocks with >> WGs (guide §1); stripes over item tiles
  int stripes = (int)std::min<int64_t>((V + 63) / 64, std::max(1, 4096 / m_tiles));
  // LDS hit-append epilogue: opt-in (REPLAY_AMD_LDSEPI=1) negative result.
  // It removes the in-loop global atomics (raw kernel -5% at K=10 shapes),
  // BUT sample-threshold noise makes a few rows ~30x hotter than expected
  // (the j-th-of-32768 order statistic has ~38% quantile std), and ANY
  // single-walk overflow of the 8-entry LDS list marks the row for the
  // exact-rescore fallback: at the K=100 bench config that is ~34 rescored
  // rows and a 2x END-TO-END regression (12.2 vs 6.4 ms).  The direct
  // global append tolerates hot rows (they only rescore when the TOTAL
  // exceeds capacity), so it stays the default.
  static const bool ldsepi_env = std::getenv("REPLAY_AMD_LDSEPI") != nullptr;
  bool use_ldsepi = ldsepi_env && (int64_t)capacity <= (int64_t)20 * stripes;
  if (variant != nullptr && variant[0] == 'w') use_ldsepi = false;  // A/B
  static const bool dbg = std::getenv("REPLAY_AMD_DEBUG") != nullptr;
  if (dbg) {
    fprintf(stderr, "[stg] E=%ld M=%ld cap=%ld stripes=%d ldsepi=%d variant=%s\n", (long)E,
            (long)M, (long)capacity, stripes, (int)use_ldsepi, variant ? variant : "-");
  }
  dim3 grid(m_tiles, stripes);
  auto stream = at::cuda::getCurrentHIPStream();
#define LAUNCH_STG(EE)                                                                   \
  hipLaunchKernelGGL((scored_topk_gemm_kernel<EE, (EE <= 128)>), grid, dim3(256),        \
                     (EE <= 128) ? 0 : (size_t)256 * EE * 2, stream,                     \
                     reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),              \
                     reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),              \
                     thr.data_ptr<float>(), out_vals.data_ptr<float>(),                  \
                     out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,              \
                     (int)capacity)
  if (E == 64 && legacy) {
    LAUNCH_STG(64);
  } else if (E == 128 && legacy) {
    LAUNCH_STG(128);
  } else if (E == 64) {
    if (use_ldsepi) {
      const size_t lds64 = 3 * 64 * (64 * 2 + 32) + 8 * (128 * 4 + 128 * 8 * 8);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<64, 8>), grid, dim3(512), lds64, stream,
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else {
      const size_t lds64 = 3 * 64 * (64 * 2 + 32);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<64, 8, 3, 0, 0, 1, 1, 0>), grid, dim3(512),
                         lds64, stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    }
  } else if (E == 128) {
    if (use_ldsepi) {
      const size_t lds128 = 3 * 64 * (128 * 2 + 32) + 8 * (64 * 4 + 64 * 8 * 8);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<128, 4>), grid, dim3(512), lds128, stream,
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else {
      const size_t lds128 = 3 * 64 * (128 * 2 + 32);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<128, 4, 3, 0, 0, 1, 1, 0>), grid,
                         dim3(512), lds128, stream,
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    }
  } else if (E == 256 && v6) {
    const size_t lds_v6 = 2 * 64 * (256 * 2 + 32);
    hipLaunchKernelGGL((scored_topk_gemm_kernel_v6<256>), grid, dim3(512), lds_v6, stream,
                       reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                       thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                       out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                       (int)capacity);
  } else if (E == 256) {
    if (variant != nullptr && variant[0] == '2') {
      // v2: 8-wave, resident-A, LDS-free, per-wave B ring (A/B reference)
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v2<256, 2>), grid, dim3(512), 0, stream,
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else if (variant != nullptr && variant[0] == '3') {
      // v3: glds double-buffer + __syncthreads (A/B reference)
      const size_t lds_v3 = 2 * 64 * (256 * 2 + 32);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v3<256>), grid, dim3(512), lds_v3, stream,
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else if (v4_mf == 2) {
      // v4 at the narrow 256-row M-tile (A/B reference)
      const size_t lds_v4 = 3 * 64 * (256 * 2 + 32);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 2, 3, 0, 0, 1, 1, 0>), grid,
                         dim3(512), lds_v4, stream,
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else if (variant != nullptr &&
               (variant[0] == '9' || variant[0] == 'a' || variant[0] == 'b')) {
      const size_t lds_ab = 3 * 64 * (256 * 2 + 32);
      if (variant[0] == '9') {
        hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 3, 3, 0, 1, 1, 0>), grid, dim3(512), lds_ab,
                           stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                           reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                           thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                           out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                           (int)capacity);
      } else if (variant[0] == 'a') {
        hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 3, 4, 0, 1, 1, 0>), grid, dim3(512), lds_ab,
                           stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                           reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                           thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                           out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                           (int)capacity);
      } else {
        hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 3, 5, 0, 1, 1, 0>), grid, dim3(512), lds_ab,
                           stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                           reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                           thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                           out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                           (int)capacity);
      }
    } else if (variant != nullptr && (variant[0] == '7' || variant[0] == '8')) {
      // ablation probes: 7 = no staging after the prologue (compute+barrier
      // only, WRONG RESULTS), 8 = no MFMA/epilogue (DMA+barrier only)
      const size_t lds_ab = 3 * 64 * (256 * 2 + 32);
      if (variant[0] == '7') {
        hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 3, 1, 0, 1, 1, 0>), grid, dim3(512), lds_ab,
                           stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                           reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                           thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                           out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                           (int)capacity);
      } else {
        hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 3, 2, 0, 1, 1, 0>), grid, dim3(512), lds_ab,
                           stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                           reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                           thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                           out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                           (int)capacity);
      }
    } else if (variant != nullptr && variant[0] == 'z') {
      // spread staging + 4-deep buffer ring (2 tiles of DMA in flight)
      const size_t lds_z = 4 * 64 * (256 * 2 + 32);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 4, 0, 0, 1, 1, 0>), grid, dim3(512),
                         lds_z, stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else if (variant != nullptr && variant[0] == 'y') {
      // A/B reference: end-burst staging (no per-f-column spread)
      const size_t lds_y = 3 * 64 * (256 * 2 + 32) + 8 * (64 * 4 + 64 * 8 * 8);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 3, 0, 0, 1, 0>), grid, dim3(512),
                         lds_y, stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else if (variant != nullptr && variant[0] == 'x') {
      // A/B reference: XCD-aware remap DISABLED
      const size_t lds_x = 3 * 64 * (256 * 2 + 32) + 8 * (64 * 4 + 64 * 8 * 8);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 3, 0, 0, 0>), grid, dim3(512),
                         lds_x, stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else if (variant != nullptr && variant[0] == 'p') {
      // loader-specialized staging: 2 waves carry the whole DMA stream
      const size_t lds_p = 3 * 64 * (256 * 2 + 32);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 3, 0, 2, 1, 1, 0>), grid, dim3(512), lds_p,
                         stream, reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else if (variant != nullptr && variant[0] == '6') {
      // v4 with a 4-deep buffer ring (3 tiles of DMA in flight)
      const size_t lds_v46 = 4 * 64 * (256 * 2 + 32);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 4, 0, 0, 1, 1, 0>), grid, dim3(512), lds_v46,
                         stream,
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else if (use_ldsepi) {
      // v4: glds triple-buffer + raw barrier + counted vmcnt, 512-row
      // M-tile, LDS hit-append epilogue
      const size_t lds_v4 = 3 * 64 * (256 * 2 + 32) + 8 * (64 * 4 + 64 * 8 * 8);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4>), grid, dim3(512), lds_v4, stream,
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    } else {
      // long-walk shapes (few stripes): direct global hit append
      const size_t lds_v4 = 3 * 64 * (256 * 2 + 32);
      hipLaunchKernelGGL((scored_topk_gemm_kernel_v4<256, 4, 3, 0, 0, 1, 1, 0>), grid,
                         dim3(512), lds_v4, stream,
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         thr.data_ptr<float>(), out_vals.data_ptr<float>(),
                         out_idx.data_ptr<int>(), counts.data_ptr<int>(), M, V,
                         (int)capacity);
    }
  } else {
    TORCH_CHECK(false, "scored_topk_gemm supports E in {64, 128, 256}");
  }
#undef LAUNCH_STG
  return {out_vals, out_idx, counts};
}
