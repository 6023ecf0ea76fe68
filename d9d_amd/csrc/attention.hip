// CDNA4 (gfx950) flash attention: forward + backward, bf16, causal, GQA, LSE.
//
// Replaces the reference's external flash-attn wheels
// (d9d/kernel/flash_attn/function.py). MI355X-first design:
//   * MFMA v_mfma_f32_16x16x32_bf16 tiles, wave64.
//   * Forward: workgroup = 4 waves x 32 q-rows (two 16-row m-tiles per wave,
//     q block 128), KV tiles of 64 staged with the T14 issue-early/write-late
//     split — K row-major with an XOR swizzle (conflict-free ds_read_b128
//     B-fragments), V transposed with vectorized 8-byte LDS writes. Online
//     softmax in fp32; s_setprio(1) around MFMA clusters; LSE written for
//     context-parallel merging.
//   * Backward: FA2 decomposition; workgroup = 8 waves owning a 128-row KV
//     tile, looping q tiles (Q/dO prefetched into registers under the MFMAs,
//     transposed images derived LDS-to-LDS); dK/dV accumulate in registers,
//     dQ via fp32 global atomics; delta = rowsum(dO*O) precomputed.
//
// Layouts: q (B,Sq,Hq,D), k/v (B,Skv,Hkv,D), out (B,Sq,Hq,D), lse (B,Hq,Sq).
// D (head_dim) templated in {32, 64, 96, 128}; host pads other sizes.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace d9d {

typedef __bf16 bf16_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

D9D_DEVICE f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// XOR swizzles keep ds_read_b128 bank-conflict-free (guide G4). The XOR must
// stay inside the LDS row: row-major [64][D] tiles (D*2-byte rows) use mask
// kSwzRM(D); transposed [D][64] tiles (128-byte rows) use mask 7.
template <int D>
constexpr int swz_rm_mask() {
  return D == 128 ? 15 : (D >= 64 ? 7 : 3);
}

union ushort2_t {  // 4 bf16 lanes packed for one 8-byte LDS store
  uint64_t u;
  ushort s[4];
};

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4_attn;
#define bf16x4 bf16x4_attn

D9D_DEVICE ushort bf16_bits(bf16_t v) {
  union { bf16_t b; ushort u; } cv;
  cv.b = v;
  return cv.u;
}

// Transpose a swizzled row-major [64][D] LDS image into a [D][64] LDS image
// (128-byte rows, ((d&7)<<4) swizzle) using 4B LDS reads + 8B LDS writes.
template <int D, int BLOCK = 256>
D9D_DEVICE void transpose_lds_tile(bf16_t* dst, const bf16_t* src_rm, int tid) {
  // Work mapping: consecutive threads take consecutive ROW-quads of the
  // same d-pair, so the two b64 writes of a 16-thread group land on one
  // [d][64] row contiguously (128 B span, conflict-free). The b32 source
  // reads pay a 4-way conflict instead of the old mapping's 8-way b64
  // write conflict (rows 256 B apart XOR-spread only 4 ways per group).
  for (int idx = tid; idx < (64 / 4) * (D / 2); idx += BLOCK) {
    const int d0 = (idx / 16) * 2;
    const int rb = (idx % 16) * 4;
    ushort2_t c0, c1;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row = rb + i;
      const int byte = (d0 * 2) ^ ((row & swz_rm_mask<D>()) << 4);
      const uint32_t pair = *reinterpret_cast<const uint32_t*>(
          reinterpret_cast<const char*>(src_rm) + row * (D * 2) + byte);
      c0.s[i] = (ushort)(pair & 0xffffu);
      c1.s[i] = (ushort)(pair >> 16);
    }
    const int byte0 = (rb * 2) ^ ((d0 & 7) << 4);
    const int byte1 = (rb * 2) ^ (((d0 + 1) & 7) << 4);
    *reinterpret_cast<uint64_t*>(
        reinterpret_cast<char*>(dst) + d0 * 128 + byte0) = c0.u;
    *reinterpret_cast<uint64_t*>(
        reinterpret_cast<char*>(dst) + (d0 + 1) * 128 + byte1) = c1.u;
  }
}

// Stage a (64 rows x D cols) global tile TRANSPOSED into a [D][64] LDS image
// (128-byte rows, ((d&7)<<4) XOR swizzle) with vectorized 8B LDS writes:
// each op covers a (4 rows x 2 cols) block.
template <int D, int ROWS = 64, int BLOCK = 256>
D9D_DEVICE void stage_transposed_tile(
    bf16_t* dst, const bf16_t* src, int64_t row_stride,
    int row0, int row_clamp, int tid) {
  for (int idx = tid; idx < (ROWS / 4) * (D / 2); idx += BLOCK) {
    const int d0 = (idx % (D / 2)) * 2;
    const int rb = (idx / (D / 2)) * 4;
    ushort2_t c0, c1;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int g_row = min(row0 + rb + i, row_clamp);
      const uint32_t pair = *reinterpret_cast<const uint32_t*>(
          src + (int64_t)g_row * row_stride + d0);
      c0.s[i] = (ushort)(pair & 0xffffu);
      c1.s[i] = (ushort)(pair >> 16);
    }
    const int byte0 = (rb * 2) ^ ((d0 & 7) << 4);
    const int byte1 = (rb * 2) ^ (((d0 + 1) & 7) << 4);
    *reinterpret_cast<uint64_t*>(
        reinterpret_cast<char*>(dst) + d0 * (ROWS * 2) + byte0) = c0.u;
    *reinterpret_cast<uint64_t*>(
        reinterpret_cast<char*>(dst) + (d0 + 1) * (ROWS * 2) + byte1) = c1.u;
  }
}

// Fragment maps for v_mfma_f32_16x16x32_bf16 (M=N=16, K=32):
//   A[m][k]: lane l holds m = l&15, k = (l>>4)*8 + j   (j = 0..7)
//   B[k][n]: lane l holds n = l&15, k = (l>>4)*8 + j
//   C[m][n]: lane l, reg r: m = (l>>4)*4 + r, n = l&15
// (verified on-device by mfma_selfcheck below).

constexpr int kQBlk = 64;   // q rows per workgroup (16 per wave)
constexpr int kKvBlk = 64;  // kv rows per tile
constexpr float kLog2e = 1.44269504088896340736f;

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------

template <int D>
__global__ __launch_bounds__(256, 2) void flash_fwd_kernel(
    const bf16_t* __restrict__ q,   // (B,Sq,Hq,D)
    const bf16_t* __restrict__ k,   // (B,Skv,Hkv,D)
    const bf16_t* __restrict__ v,   // (B,Skv,Hkv,D)
    bf16_t* __restrict__ out,       // (B,Sq,Hq,D)
    float* __restrict__ lse,        // (B,Hq,Sq)
    const float* __restrict__ sinks,  // (Hq,) raw sink logits, or nullptr
    const int* __restrict__ cu_q,     // (nseq+1,) packed-seq bounds, or nullptr
    const int* __restrict__ cu_k,
    const int* __restrict__ qtile_pref,  // (nseq+1,) prefix of ceil(len_q/128)
    int nseq,
    int B, int Sq, int Skv, int Hq, int Hkv,
    float scale, int causal, int window_left, int q_offset) {
  constexpr int kNT = D / 16;   // n-tiles over head dim
  constexpr int kKS = D / 32;   // k-steps over head dim
  constexpr int kRowBytes = D * 2;
  constexpr int kQB = 128;      // q rows per workgroup: 32 per wave (2 m-tiles)

  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* k_lds = reinterpret_cast<bf16_t*>(smem);            // [64][D] swizzled
  bf16_t* vt_lds = k_lds + kKvBlk * D;                        // [D][64] transposed
  // per-wave P scratch: [4][32][kKvBlk+8]
  bf16_t* p_lds = vt_lds + D * kKvBlk;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  // Varlen (packed sequences): grid.x spans per-sequence q-tiles; binary
  // search the tile prefix for our sequence, then work in LOCAL coordinates
  // with the sequence's row ranges. Causal masking aligns the END of q with
  // the END of kv (flash-attn convention for len_q != len_kv).
  int b, h, q_tile, q_lo, Sq_loc, kv_lo, Skv_loc, causal_off;
  if (cu_q != nullptr) {
    h = blockIdx.y;
    int lo = 0, hi = nseq - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (qtile_pref[mid] <= (int)blockIdx.x) lo = mid; else hi = mid - 1;
    }
    b = 0;
    q_tile = blockIdx.x - qtile_pref[lo];
    q_lo = cu_q[lo];
    Sq_loc = cu_q[lo + 1] - q_lo;
    kv_lo = cu_k[lo];
    Skv_loc = cu_k[lo + 1] - kv_lo;
    causal_off = Skv_loc - Sq_loc + q_offset;
    if (Sq_loc <= 0 || q_tile * kQB >= Sq_loc) return;
  } else {
    // XCD-aware mapping: group one (b, h)'s q-tiles onto one XCD so the
    // KV stream they share is served from that XCD's L2 instead of every
    // XCD re-streaming it from HBM (see gmm_nt_kernel for the bijection).
    const int nwg = gridDim.x * gridDim.y;
    const int lin = blockIdx.y * gridDim.x + blockIdx.x;
    const int xcd = lin & 7;
    const int qd = nwg >> 3, rd = nwg & 7;
    const int wid = (xcd < rd ? xcd * (qd + 1) : rd * (qd + 1) + (xcd - rd) * qd)
                    + (lin >> 3);
    const int bh = wid / gridDim.x;
    b = bh / Hq;
    h = bh % Hq;
    q_tile = wid % gridDim.x;
    q_lo = 0; Sq_loc = Sq; kv_lo = 0; Skv_loc = Skv;
    causal_off = q_offset;
  }
  const int hkv = h / (Hq / Hkv);

  const int64_t q_base = ((int64_t)b * Sq * Hq + h) * D;
  const int64_t kv_base = ((int64_t)b * Skv * Hkv + hkv) * D;
  const int64_t q_row_stride = (int64_t)Hq * D;
  const int64_t kv_row_stride = (int64_t)Hkv * D;

  // ---- load this wave's 32 q rows (2 m-tiles) into A fragments -------------
  bf16x8 q_frag[2][kKS];
#pragma unroll
  for (int m = 0; m < 2; ++m) {
    const int q_row_local = q_tile * kQB + wave * 32 + m * 16 + (lane & 15);
    const int safe_row = q_lo + min(q_row_local, Sq_loc - 1);
    const bf16_t* qp = q + q_base + (int64_t)safe_row * q_row_stride;
#pragma unroll
    for (int ks = 0; ks < kKS; ++ks) {
      q_frag[m][ks] = *reinterpret_cast<const bf16x8*>(qp + ks * 32 + (lane >> 4) * 8);
    }
  }

  // A learnable "sink" is one virtual softmax column per head with logit
  // sinks[h] and no value row: seed the running max/sum with it and the
  // online softmax (and the LSE the backward reads) absorbs it for free.
  const float m_init = sinks ? sinks[h] : -1e30f;
  // l is tracked per lane and merged at the epilogue: seed the sink's
  // softmax column on ONE lane of each row group only
  const float l_init = (sinks && (threadIdx.x & 15) == 0) ? 1.f : 0.f;
  float m_run[2][4], l_run[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[m][r] = m_init;
      l_run[m][r] = l_init;
    }
  f32x4 o_acc[2][kNT];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int nt = 0; nt < kNT; ++nt) o_acc[m][nt] = {0.f, 0.f, 0.f, 0.f};

  const int q_tile_last_row = min(q_tile * kQB + kQB - 1, Sq_loc - 1);
  int kv_end = Skv_loc;
  if (causal) kv_end = min(Skv_loc, q_tile_last_row + causal_off + 1);
  const int num_kv_tiles = (kv_end + kKvBlk - 1) / kKvBlk;
  if (num_kv_tiles <= 0) return;

  // T14 split staging: next tile's K rows and V (4x2 transpose blocks) are
  // loaded into registers while the current tile's MFMAs run; LDS writes and
  // their vmcnt waits land after the compute barrier.
  bf16x8 k_reg[4];
  ushort2_t v_c0[4], v_c1[4];

  auto load_regs = [&](int kt) {
    const int kv0 = kt * kKvBlk;
#pragma unroll
    for (int it = 0; it < (kKvBlk * D) / (256 * 8); ++it) {
      const int idx = (threadIdx.x + it * 256) * 8;
      const int row = idx / D;
      const int col = idx % D;
      const int g_row = kv_lo + min(kv0 + row, Skv_loc - 1);
      k_reg[it] = *reinterpret_cast<const bf16x8*>(
          k + kv_base + (int64_t)g_row * kv_row_stride + col);
    }
#pragma unroll
    for (int it = 0; it < (kKvBlk / 4) * (D / 2) / 256; ++it) {
      const int idx = threadIdx.x + it * 256;
      const int d0 = (idx % (D / 2)) * 2;
      const int kvb = (idx / (D / 2)) * 4;
      v_c0[it].u = v_c1[it].u = 0;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int g_row = kv_lo + min(kv0 + kvb + i, Skv_loc - 1);
        const uint32_t pair = *reinterpret_cast<const uint32_t*>(
            v + kv_base + (int64_t)g_row * kv_row_stride + d0);
        v_c0[it].s[i] = (ushort)(pair & 0xffffu);
        v_c1[it].s[i] = (ushort)(pair >> 16);
      }
    }
  };

  auto store_lds = [&]() {
#pragma unroll
    for (int it = 0; it < (kKvBlk * D) / (256 * 8); ++it) {
      const int idx = (threadIdx.x + it * 256) * 8;
      const int row = idx / D;
      const int col = idx % D;
      const int byte = (col * 2) ^ ((row & swz_rm_mask<D>()) << 4);
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(k_lds) + row * kRowBytes + byte) = k_reg[it];
    }
#pragma unroll
    for (int it = 0; it < (kKvBlk / 4) * (D / 2) / 256; ++it) {
      const int idx = threadIdx.x + it * 256;
      const int d0 = (idx % (D / 2)) * 2;
      const int kvb = (idx / (D / 2)) * 4;
      const int byte0 = (kvb * 2) ^ ((d0 & 7) << 4);
      const int byte1 = (kvb * 2) ^ (((d0 + 1) & 7) << 4);
      *reinterpret_cast<uint64_t*>(
          reinterpret_cast<char*>(vt_lds) + d0 * (kKvBlk * 2) + byte0) = v_c0[it].u;
      *reinterpret_cast<uint64_t*>(
          reinterpret_cast<char*>(vt_lds) + (d0 + 1) * (kKvBlk * 2) + byte1) = v_c1[it].u;
    }
  };

  load_regs(0);
  store_lds();
  __syncthreads();

  for (int kt = 0; kt < num_kv_tiles; ++kt) {
    const int kv0 = kt * kKvBlk;
    if (kt + 1 < num_kv_tiles) load_regs(kt + 1);  // issue early

    // per m-tile: QK^T -> softmax -> P -> PV
#pragma unroll
    for (int m = 0; m < 2; ++m) {
      f32x4 s_acc[4];
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) s_acc[nt] = {0.f, 0.f, 0.f, 0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int kv_row = nt * 16 + (lane & 15);
#pragma unroll
        for (int ks = 0; ks < kKS; ++ks) {
          const int d0 = ks * 32 + (lane >> 4) * 8;
          const int byte = (d0 * 2) ^ ((kv_row & swz_rm_mask<D>()) << 4);
          const bf16x8 kb = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<char*>(k_lds) + kv_row * kRowBytes + byte);
          s_acc[nt] = mfma16(q_frag[m][ks], kb, s_acc[nt]);
        }
      }
      __builtin_amdgcn_s_setprio(0);

      const int my_q_row = q_tile * kQB + wave * 32 + m * 16;
      float p_val[4][4];  // [nt][r]
      float m_new[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) m_new[r] = m_run[m][r];
      // Tile-level mask skip: most KV tiles sit fully below the causal
      // diagonal and inside the window for every row of this m-tile -- the
      // per-element mask chain (3 compares + select x16) only runs on the
      // diagonal/edge tiles (wave-uniform branch).
      const int row_lo = my_q_row + causal_off;
      const bool any_mask =
          (kv0 + kKvBlk > kv_end) ||
          (causal && kv0 + kKvBlk - 1 > row_lo) ||
          (window_left >= 0 && kv0 < row_lo + 15 - window_left);
      if (any_mask) {
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          const int col = kv0 + nt * 16 + (lane & 15);
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = my_q_row + (lane >> 4) * 4 + r + causal_off;
            float s = s_acc[nt][r] * scale;
            bool masked = col >= kv_end;
            if (causal) masked |= col > row;
            if (window_left >= 0) masked |= col < row - window_left;
            p_val[nt][r] = masked ? -1e30f : s;
          }
        }
      } else {
#pragma unroll
        for (int nt = 0; nt < 4; ++nt)
#pragma unroll
          for (int r = 0; r < 4; ++r) p_val[nt][r] = s_acc[nt][r] * scale;
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float mx = fmaxf(fmaxf(p_val[0][r], p_val[1][r]),
                         fmaxf(p_val[2][r], p_val[3][r]));
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) {
          mx = fmaxf(mx, __shfl_xor(mx, off, 64));
        }
        m_new[r] = fmaxf(m_new[r], mx);
      }
      // l accumulates PER LANE (each lane's own 4 columns): the row-wide
      // sum only matters at the epilogue, where one 16-lane merge replaces
      // a 4-step shuffle cascade per row per KV tile (the fwd kernel's
      // PMC showed a 10:1 VALU:MFMA instruction ratio).
      float l_add[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float acc = 0.f;
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          const float p = __builtin_amdgcn_exp2f((p_val[nt][r] - m_new[r]) * kLog2e);
          p_val[nt][r] = p;
          acc += p;
        }
        l_add[r] = acc;
      }
      // Lazy rescale: after the first few KV tiles the running max rarely
      // moves, so alpha == 1 for every row of the wave most of the time --
      // skip the 4 exp2s and kNT*4 accumulator multiplies (wave-uniform
      // branch, ~25% of the softmax VALU work).
      bool same = true;
#pragma unroll
      for (int r = 0; r < 4; ++r) same &= (m_new[r] == m_run[m][r]);
      if (__all(same)) {
#pragma unroll
        for (int r = 0; r < 4; ++r) l_run[m][r] += l_add[r];
      } else {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float alpha = __builtin_amdgcn_exp2f((m_run[m][r] - m_new[r]) * kLog2e);
          l_run[m][r] = l_run[m][r] * alpha + l_add[r];
          m_run[m][r] = m_new[r];
#pragma unroll
          for (int nt = 0; nt < kNT; ++nt) o_acc[m][nt][r] *= alpha;
        }
      }

      // P C-layout -> row-major LDS scratch (own-wave slice) -> A fragments
      {
        bf16_t* pw = p_lds + (wave * 32 + m * 16) * (kKvBlk + 8);
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = (lane >> 4) * 4 + r;
            const int col = nt * 16 + (lane & 15);
            pw[row * (kKvBlk + 8) + col] = (bf16_t)p_val[nt][r];
          }
        }
        const bf16_t* pr = pw;
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks2 = 0; ks2 < 2; ++ks2) {
          const int kv_off = ks2 * 32 + (lane >> 4) * 8;
          const bf16x8 pa = *reinterpret_cast<const bf16x8*>(
              pr + (lane & 15) * (kKvBlk + 8) + kv_off);
#pragma unroll
          for (int nt = 0; nt < kNT; ++nt) {
            const int d = nt * 16 + (lane & 15);
            const int byte = (kv_off * 2) ^ ((d & 7) << 4);
            const bf16x8 vb = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<char*>(vt_lds) + d * (kKvBlk * 2) + byte);
            o_acc[m][nt] = mfma16(pa, vb, o_acc[m][nt]);
          }
        }
        __builtin_amdgcn_s_setprio(0);
      }
    }
    __syncthreads();
    if (kt + 1 < num_kv_tiles) {
      store_lds();
      __syncthreads();
    }
  }

  // ---- epilogue: merge per-lane l across the 16 column lanes (once),
  // then O /= l; stage in LDS; coalesced 16B stores; LSE --------------------
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int r = 0; r < 4; ++r)
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        l_run[m][r] += __shfl_xor(l_run[m][r], off, 64);
  {
    bf16_t* o_lds = k_lds;  // reuse: [64][D] staged twice (two 64-row halves)
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      __syncthreads();
      // waves 0,1 hold q rows 0..63 of the tile; waves 2,3 rows 64..127
      if ((wave >> 1) == half) {
#pragma unroll
        for (int m = 0; m < 2; ++m) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float inv_l = (l_run[m][r] > 0.f) ? 1.f / l_run[m][r] : 0.f;
            const int row = (wave & 1) * 32 + m * 16 + (lane >> 4) * 4 + r;
#pragma unroll
            for (int nt = 0; nt < kNT; ++nt) {
              o_lds[row * D + nt * 16 + (lane & 15)] =
                  (bf16_t)(o_acc[m][nt][r] * inv_l);
            }
          }
        }
      }
      __syncthreads();
      constexpr int elems = 64 * D;
      for (int idx = threadIdx.x * 8; idx < elems; idx += 256 * 8) {
        const int row = idx / D;
        const int col = idx % D;
        const int rg = q_tile * kQB + half * 64 + row;
        if (rg < Sq_loc) {
          *reinterpret_cast<bf16x8*>(
              out + q_base + (int64_t)(q_lo + rg) * q_row_stride + col) =
              *reinterpret_cast<const bf16x8*>(o_lds + row * D + col);
        }
      }
    }
    if ((lane & 15) == 0) {
#pragma unroll
      for (int m = 0; m < 2; ++m) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int rg = q_tile * kQB + wave * 32 + m * 16 + (lane >> 4) * 4 + r;
          if (rg < Sq_loc) {
            // regular: (B,Hq,Sq); varlen: (Hq,total) with b == 0
            lse[((int64_t)b * Hq + h) * Sq + q_lo + rg] =
                m_run[m][r] + __logf(fmaxf(l_run[m][r], 1e-30f));
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Backward
// ---------------------------------------------------------------------------

// delta[b,h,q] = sum_d dO * O  (fp32)
__global__ void attn_delta_kernel(
    const bf16_t* __restrict__ dout,  // (B,Sq,Hq,D)
    const bf16_t* __restrict__ out,
    float* __restrict__ delta,        // (B,Hq,Sq)
    int B, int Sq, int Hq, int D) {
  const int64_t row = blockIdx.x;  // over B*Sq*Hq
  if (row >= (int64_t)B * Sq * Hq) return;
  const int h = row % Hq;
  const int s = (row / Hq) % Sq;
  const int b = row / ((int64_t)Hq * Sq);
  const bf16_t* dp = dout + row * D;
  const bf16_t* op = out + row * D;
  float acc = 0.f;
  for (int i = threadIdx.x; i < D; i += 64) {
    acc += (float)dp[i] * (float)op[i];
  }
  acc = wave_reduce_sum(acc);
  if (threadIdx.x == 0) {
    delta[((int64_t)b * Hq + h) * Sq + s] = acc;
  }
}

constexpr int kBwdKv = 128;      // kv rows per block (8 waves x 16)
constexpr int kBwdThreads = 512;

template <int D>
__global__ __launch_bounds__(512, 1) void flash_bwd_kernel(
    const bf16_t* __restrict__ q,
    const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v,
    const bf16_t* __restrict__ dout,
    const float* __restrict__ lse,    // (B,Hq,Sq)
    const float* __restrict__ delta,  // (B,Hq,Sq)
    float* __restrict__ dq,           // (B,Sq,Hq,D) fp32 accum
    float* __restrict__ dk,           // (B,Skv,Hkv,D) fp32 accum
    float* __restrict__ dv,           // (B,Skv,Hkv,D) fp32 accum
    const int* __restrict__ cu_q,     // packed-seq bounds, or nullptr
    const int* __restrict__ cu_k,
    const int* __restrict__ kvtile_pref,  // prefix of ceil(len_k/kBwdKv)
    int nseq,
    int B, int Sq, int Skv, int Hq, int Hkv,
    float scale, int causal, int window_left, int q_offset) {
  constexpr int kNT = D / 16;
  constexpr int kKS = D / 32;
  constexpr int kRowBytes = D * 2;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // Separate buffers for every staged operand: the three MFMA phases
  // (dV, dK, dQ) then run back-to-back with NO barrier between them, and the
  // whole q-tile iteration needs 4 __syncthreads instead of 8 (PMC showed
  // ~50% of bwd wave cycles parked on barriers with the shared-buffer
  // schedule). 149 KB LDS at D=128 -- one workgroup per CU, which this
  // kernel is at anyway.
  bf16_t* q_lds = reinterpret_cast<bf16_t*>(smem);   // [64][D] swizzled rows
  bf16_t* do_lds = q_lds + kQBlk * D;                // [64][D]
  bf16_t* tdo_lds = do_lds + kQBlk * D;              // [D][64]: dO^T
  bf16_t* tq_lds = tdo_lds + D * kQBlk;              // [D][64]: Q^T
  bf16_t* kt_lds = tq_lds + D * kQBlk;               // [D][kBwdKv]
  // P and dS live in ONE q-major blocked image each: [4 q][16 kv] 128-B
  // tr16 blocks at (q>>2, kv>>4), qb stride padded +32 B so the packed
  // b64 stores of 4 kv values per (q, nt) are bank-spread. dV/dK read
  // their kv-major A fragments via ds_read_b64_tr_b16 pairs; dQ reads its
  // q-major fragments as plain 16-B rows of the SAME dS image — the old
  // third (q-major) copy and 48 scalar stores per wave are gone.
  constexpr int kXkvB = kBwdKv / 16;                 // kv blocks per q-row
  constexpr int kXqStride = kXkvB * 128 + 32;        // bytes per q-block row
  char* x1_lds = reinterpret_cast<char*>(kt_lds + D * kBwdKv);  // P image
  char* x2_lds = x1_lds + (kQBlk / 4) * kXqStride;              // dS image

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;   // 8 waves x 16 kv rows = 128 kv/block

  // Varlen: grid.x spans per-sequence kv-tiles; local coordinates within
  // the sequence, causal offset aligns q end to kv end (see fwd kernel).
  int b, h, kv_tile, q_lo, Sq_loc, kv_lo, Skv_loc, causal_off;
  if (cu_q != nullptr) {
    h = blockIdx.y;
    int lo = 0, hi = nseq - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (kvtile_pref[mid] <= (int)blockIdx.x) lo = mid; else hi = mid - 1;
    }
    b = 0;
    kv_tile = blockIdx.x - kvtile_pref[lo];
    q_lo = cu_q[lo];
    Sq_loc = cu_q[lo + 1] - q_lo;
    kv_lo = cu_k[lo];
    Skv_loc = cu_k[lo + 1] - kv_lo;
    causal_off = Skv_loc - Sq_loc + q_offset;
    if (Skv_loc <= 0 || Sq_loc <= 0 || kv_tile * kBwdKv >= Skv_loc) return;
  } else {
    // XCD-aware mapping (see fwd): one (b, h)'s kv-tiles share its Q/dO/lse
    // streams through a single XCD's L2.
    const int nwg = gridDim.x * gridDim.y;
    const int lin = blockIdx.y * gridDim.x + blockIdx.x;
    const int xcd = lin & 7;
    const int qd = nwg >> 3, rd = nwg & 7;
    const int wid = (xcd < rd ? xcd * (qd + 1) : rd * (qd + 1) + (xcd - rd) * qd)
                    + (lin >> 3);
    const int bh = wid / gridDim.x;
    b = bh / Hq;
    h = bh % Hq;
    kv_tile = wid % gridDim.x;
    q_lo = 0; Sq_loc = Sq; kv_lo = 0; Skv_loc = Skv;
    causal_off = q_offset;
  }
  const int hkv = h / (Hq / Hkv);

  const int64_t q_base = ((int64_t)b * Sq * Hq + h) * D;
  const int64_t kv_base = ((int64_t)b * Skv * Hkv + hkv) * D;
  const int64_t q_row_stride = (int64_t)Hq * D;
  const int64_t kv_row_stride = (int64_t)Hkv * D;
  const float* lse_row = lse + ((int64_t)b * Hq + h) * Sq + q_lo;
  const float* delta_row = delta + ((int64_t)b * Hq + h) * Sq + q_lo;

  const int kv0 = kv_tile * kBwdKv;

  // This wave's 16 kv rows: K and V A-fragments in registers.
  const int kv_row_local = lane & 15;
  const int kv_row_global =
      kv_lo + min(kv0 + wave * 16 + kv_row_local, Skv_loc - 1);
  bf16x8 k_frag[kKS], v_frag[kKS];
  {
    const bf16_t* kp = k + kv_base + (int64_t)kv_row_global * kv_row_stride;
    const bf16_t* vp = v + kv_base + (int64_t)kv_row_global * kv_row_stride;
#pragma unroll
    for (int ks = 0; ks < kKS; ++ks) {
      const int d0 = ks * 32 + (lane >> 4) * 8;
      k_frag[ks] = *reinterpret_cast<const bf16x8*>(kp + d0);
      v_frag[ks] = *reinterpret_cast<const bf16x8*>(vp + d0);
    }
  }

  // Stage K^T once (for dQ's B fragments).
  stage_transposed_tile<D, kBwdKv, kBwdThreads>(
      kt_lds, k + kv_base + (int64_t)kv_lo * kv_row_stride, kv_row_stride, kv0,
      Skv_loc - 1, threadIdx.x);

  f32x4 dk_acc[kNT], dv_acc[kNT];
#pragma unroll
  for (int nt = 0; nt < kNT; ++nt) {
    dk_acc[nt] = {0.f, 0.f, 0.f, 0.f};
    dv_acc[nt] = {0.f, 0.f, 0.f, 0.f};
  }

  int q_start = 0;
  if (causal) {
    // first q tile whose last aligned position reaches kv0
    q_start = (max(kv0 - causal_off, 0) / kQBlk) * kQBlk;
  }
  if (q_start >= Sq_loc) return;

  // T14: each q-tile's Q/dO rows are prefetched into registers while the
  // previous tile's MFMAs run; the transposed images are derived from the
  // row-major LDS copies (cheap ds_reads) instead of re-reading global.
  constexpr int kQdoRegs =
      (kQBlk * D + kBwdThreads * 8 - 1) / (kBwdThreads * 8);  // >= 1 (D <= 128)
  bf16x8 q_reg[kQdoRegs], do_reg[kQdoRegs];

  auto load_qdo_regs = [&](int qt_next) {
#pragma unroll
    for (int it = 0; it < kQdoRegs; ++it) {
      const int idx = (threadIdx.x + it * kBwdThreads) * 8;
      if (idx >= kQBlk * D) break;
      const int row = idx / D;
      const int col = idx % D;
      const int g_row = q_lo + min(qt_next + row, Sq_loc - 1);
      q_reg[it] = *reinterpret_cast<const bf16x8*>(
          q + q_base + (int64_t)g_row * q_row_stride + col);
      do_reg[it] = *reinterpret_cast<const bf16x8*>(
          dout + q_base + (int64_t)g_row * q_row_stride + col);
    }
  };
  auto store_qdo_lds = [&]() {
#pragma unroll
    for (int it = 0; it < kQdoRegs; ++it) {
      const int idx = (threadIdx.x + it * kBwdThreads) * 8;
      if (idx >= kQBlk * D) break;
      const int row = idx / D;
      const int col = idx % D;
      const int byte = (col * 2) ^ ((row & swz_rm_mask<D>()) << 4);
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(q_lds) + row * kRowBytes + byte) = q_reg[it];
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(do_lds) + row * kRowBytes + byte) = do_reg[it];
    }
  };

  load_qdo_regs(q_start);
  store_qdo_lds();
  __syncthreads();

  for (int qt = q_start; qt < Sq_loc; qt += kQBlk) {
    if (qt + kQBlk < Sq_loc) load_qdo_regs(qt + kQBlk);  // issue early
    // dO^T and Q^T images for the dV/dK B fragments (cheap LDS-to-LDS).
    // NO barrier here: tdo/tq are first read after the x-stage barrier
    // below, which also publishes these writes (2 barriers per q-tile
    // instead of 4 — the PMC wave-parked share was the top bwd cost).
    transpose_lds_tile<D, kBwdThreads>(tdo_lds, do_lds, threadIdx.x);
    transpose_lds_tile<D, kBwdThreads>(tq_lds, q_lds, threadIdx.x);

    // ---- S^T = K @ Q^T (per wave: 16 kv x 64 q) ----------------------------
    f32x4 st_acc[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) st_acc[nt] = {0.f, 0.f, 0.f, 0.f};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int q_row = nt * 16 + (lane & 15);
#pragma unroll
      for (int ks = 0; ks < kKS; ++ks) {
        const int d0 = ks * 32 + (lane >> 4) * 8;
        const int byte = (d0 * 2) ^ ((q_row & swz_rm_mask<D>()) << 4);
        const bf16x8 qb = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<char*>(q_lds) + q_row * kRowBytes + byte);
        st_acc[nt] = mfma16(k_frag[ks], qb, st_acc[nt]);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- P^T = exp(S^T * scale - lse[q]); masked entries 0 -----------------
    // Tile-level mask skip (see fwd): interior tiles take the select-free
    // path, only diagonal/edge tiles run the per-element mask chain.
    const bool any_mask =
        (kv0 + kBwdKv > Skv_loc) || (qt + kQBlk > Sq_loc) ||
        (causal && kv0 + kBwdKv - 1 > qt + causal_off) ||
        (window_left >= 0 &&
         kv0 < qt + kQBlk - 1 + causal_off - window_left);
    float pt_val[4][4];  // [nt][r]
    if (any_mask) {
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int q_glob = qt + nt * 16 + (lane & 15);
        const float l = (q_glob < Sq_loc) ? lse_row[min(q_glob, Sq_loc - 1)] : 1e30f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kv_glob = kv0 + wave * 16 + (lane >> 4) * 4 + r;
          float s = st_acc[nt][r] * scale;
          bool masked = (kv_glob >= Skv_loc) || (q_glob >= Sq_loc);
          if (causal) masked |= kv_glob > q_glob + causal_off;
          if (window_left >= 0) masked |= kv_glob < q_glob + causal_off - window_left;
          pt_val[nt][r] =
              masked ? 0.f : __builtin_amdgcn_exp2f((s - l) * kLog2e);
        }
      }
    } else {
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        // clamp even on the fast path: the compiler may speculate this load
        // across the branch, and edge tiles would read past the lse buffer
        const float l = lse_row[min(qt + nt * 16 + (lane & 15), Sq_loc - 1)];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          pt_val[nt][r] =
              __builtin_amdgcn_exp2f((st_acc[nt][r] * scale - l) * kLog2e);
        }
      }
    }

    // ---- dP^T = V @ dO^T ---------------------------------------------------
    f32x4 dpt_acc[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) dpt_acc[nt] = {0.f, 0.f, 0.f, 0.f};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int q_row = nt * 16 + (lane & 15);
#pragma unroll
      for (int ks = 0; ks < kKS; ++ks) {
        const int d0 = ks * 32 + (lane >> 4) * 8;
        const int byte = (d0 * 2) ^ ((q_row & swz_rm_mask<D>()) << 4);
        const bf16x8 db = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<char*>(do_lds) + q_row * kRowBytes + byte);
        dpt_acc[nt] = mfma16(v_frag[ks], db, dpt_acc[nt]);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- dS^T = P^T * (dP^T - delta[q]) * scale ---------------------------
    float dst_val[4][4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int q_glob = qt + nt * 16 + (lane & 15);
      const float dlt = (q_glob < Sq_loc) ? delta_row[min(q_glob, Sq_loc - 1)] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        dst_val[nt][r] = pt_val[nt][r] * (dpt_acc[nt][r] - dlt) * scale;
      }
    }

    // ---- stage P and dS into the blocked images, ONE barrier --------------
    // value (q, kv) at qb = q>>2 row, block kv>>4, inner (q&3)*32 + (kv&15)*2;
    // the 4 r-values (4 consecutive kv at fixed q) pack into one b64 store.
    {
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int q_l = nt * 16 + (lane & 15);
        const int kv_l = wave * 16 + (lane >> 4) * 4;
        const size_t byte = (size_t)(q_l >> 2) * kXqStride +
                            (kv_l >> 4) * 128 + (q_l & 3) * 32 +
                            (kv_l & 15) * 2;
        ushort2_t p4, d4;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          p4.s[r] = bf16_bits((bf16_t)pt_val[nt][r]);
          d4.s[r] = bf16_bits((bf16_t)dst_val[nt][r]);
        }
        *reinterpret_cast<uint64_t*>(x1_lds + byte) = p4.u;
        *reinterpret_cast<uint64_t*>(x2_lds + byte) = d4.u;
      }
      __syncthreads();
    }

    // ---- dV += P^T @ dO; dK += dS^T @ Q: back-to-back, no barrier ----------
    {
      // A fragment (m = kv row wave*16 + lane&15, k = 8 q at q_off) = two
      // ds_read_b64_tr_b16 of the [4 q][16 kv] blocks (qb, wave)
      auto xfrag = [&](const char* img, int q_off) -> bf16x8 {
        const size_t b0 = (size_t)(q_off >> 2) * kXqStride + wave * 128 +
                          (lane & 15) * 8;
        bf16x4 lo4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (__attribute__((address_space(3))) bf16x4*)(img + b0));
        bf16x4 hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (__attribute__((address_space(3))) bf16x4*)(img + b0 + kXqStride));
        bf16x8 f;
#pragma unroll
        for (int t2 = 0; t2 < 4; ++t2) { f[t2] = lo4[t2]; f[4 + t2] = hi4[t2]; }
        return f;
      };
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks2 = 0; ks2 < 2; ++ks2) {
        const int q_off = ks2 * 32 + (lane >> 4) * 8;
        const bf16x8 pa = xfrag(x1_lds, q_off);
        const bf16x8 da = xfrag(x2_lds, q_off);
#pragma unroll
        for (int nt = 0; nt < kNT; ++nt) {
          const int d = nt * 16 + (lane & 15);
          const int byte = (q_off * 2) ^ ((d & 7) << 4);
          const bf16x8 dob = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<char*>(tdo_lds) + d * (kQBlk * 2) + byte);
          const bf16x8 qb = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<char*>(tq_lds) + d * (kQBlk * 2) + byte);
          dv_acc[nt] = mfma16(pa, dob, dv_acc[nt]);
          dk_acc[nt] = mfma16(da, qb, dk_acc[nt]);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }

    // ---- dQ partial = dS @ K; atomicAdd (A = dS q-major via x3) -----------
    {
      // ALL 8 waves: wave & 3 picks the 16-row q m-tile, wave >> 2 the
      // d-half -- the old waves-0-3-only split parked half the block
      // through an entire MFMA phase.
      constexpr int kNTH = kNT > 1 ? kNT / 2 : 1;  // nt per d-half
      const int q_mt = wave & 3;
      const int nt0 = kNT > 1 ? (wave >> 2) * kNTH : 0;
      if (kNT > 1 || wave < 4) {
        f32x4 dq_acc[kNTH];
#pragma unroll
        for (int nt = 0; nt < kNTH; ++nt) dq_acc[nt] = {0.f, 0.f, 0.f, 0.f};
        const int q_row = q_mt * 16 + (lane & 15);
        const size_t qbase = (size_t)(q_row >> 2) * kXqStride + (q_row & 3) * 32;
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks2 = 0; ks2 < kBwdKv / 32; ++ks2) {
          const int kv_off = ks2 * 32 + (lane >> 4) * 8;
          const bf16x8 da = *reinterpret_cast<const bf16x8*>(
              x2_lds + qbase + (kv_off >> 4) * 128 + (kv_off & 15) * 2);
#pragma unroll
          for (int nt = 0; nt < kNTH; ++nt) {
            const int d = (nt0 + nt) * 16 + (lane & 15);
            const int byte = (kv_off * 2) ^ ((d & 7) << 4);
            const bf16x8 kb = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<char*>(kt_lds) + d * (kBwdKv * 2) + byte);
            dq_acc[nt] = mfma16(da, kb, dq_acc[nt]);
          }
        }
        __builtin_amdgcn_s_setprio(0);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int q_glob = qt + q_mt * 16 + (lane >> 4) * 4 + r;
          if (q_glob < Sq_loc) {
#pragma unroll
            for (int nt = 0; nt < kNTH; ++nt) {
              atomicAdd(dq + q_base +
                            (int64_t)(q_lo + q_glob) * q_row_stride +
                            (nt0 + nt) * 16 + (lane & 15),
                        dq_acc[nt][r]);
            }
          }
        }
      }
    }
    // The Q/dO store overlaps the dV/dK/dQ MFMA phases above: nothing in
    // them reads q_lds/do_lds (their consumers — S^T, dP^T and the
    // transposes — all ran before the x-stage barrier), so the write only
    // needs the end-of-iteration barrier to publish for the next tile.
    if (qt + kQBlk < Sq_loc) {
      store_qdo_lds();
    }
    __syncthreads();
  }

  // ---- flush dK, dV (atomicAdd: GQA groups and padded tiles overlap) -------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kv_glob = kv0 + wave * 16 + (lane >> 4) * 4 + r;
    if (kv_glob < Skv_loc) {
#pragma unroll
      for (int nt = 0; nt < kNT; ++nt) {
        const int d = nt * 16 + (lane & 15);
        atomicAdd(dk + kv_base + (int64_t)(kv_lo + kv_glob) * kv_row_stride + d,
                  dk_acc[nt][r]);
        atomicAdd(dv + kv_base + (int64_t)(kv_lo + kv_glob) * kv_row_stride + d,
                  dv_acc[nt][r]);
      }
    }
  }
}

// flash bwd v2 — the occupancy-first geometry (opt-in: D9D_FLASH_BWD_V2=1).
// MEASURED RESULT: 176 TF/s vs the default kernel's 326 at B8 S4096 H16/4
// D128 causal — occupancy does NOT pay here. The default kernel is pinned
// at 2 waves/SIMD (212 VGPR, 127 KB LDS) and this variant reaches 3
// waves/SIMD x 3 blocks/CU, but the price of fitting 168 VGPRs / 51 KB is
// 4x K/V fragment re-reads (streamed from L2 per n-tile), a 16-KB K-tile
// re-stage per q-iteration, double the Q/dO traffic (kv-64 tiles instead
// of kv-128) and atomic dQ — memory work grows faster than the extra
// latency hiding recovers. Kept as a correct (15/15 parity) reference
// point for the occupancy/traffic trade-off; see docs/perf_notes.md.
// Original design notes:
//   * kv tile 64, ONE 4-wave (256-thread) workgroup per tile, ~49 KB LDS
//     and a <=168-VGPR budget -> 3 workgroups per CU, 3 waves per SIMD.
//   * Q and dO live in ONE [4 q][16 d] 128-B-blocked image each: the
//     S^T/dP^T B fragments are plain 16-B row reads, the dV/dK B
//     fragments are ds_read_b64_tr_b16 pairs of the same blocks — the
//     LDS->LDS transposed copies (32 KB + a pass per tile) are gone.
//   * dQ's K^T fragments read straight from global K (the 16-KB tile is
//     L2-resident; each fragment row is a 32-B coalesced lane group) —
//     the 16-KB K^T image is gone.
//   * P/dS stage once into q-major tr16 block images (as v1).
template <int D>
__global__ __launch_bounds__(256, 3) void flash_bwd_kernel_v2(
    const bf16_t* __restrict__ q,
    const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v,
    const bf16_t* __restrict__ dout,
    const float* __restrict__ lse,    // (B,Hq,Sq)
    const float* __restrict__ delta,  // (B,Hq,Sq)
    float* __restrict__ dq,           // (B,Sq,Hq,D) fp32 accum
    float* __restrict__ dk,           // (B,Skv,Hkv,D) fp32 accum
    float* __restrict__ dv,           // (B,Skv,Hkv,D) fp32 accum
    const int* __restrict__ cu_q,     // packed-seq bounds, or nullptr
    const int* __restrict__ cu_k,
    const int* __restrict__ kvtile_pref,  // prefix of ceil(len_k/64)
    int nseq,
    int B, int Sq, int Skv, int Hq, int Hkv,
    float scale, int causal, int window_left, int q_offset) {
  constexpr int kNT = D / 16;
  constexpr int kKS = D / 32;
  constexpr int kKv = 64;                   // kv rows per workgroup
  constexpr int kDB = D / 16;               // d blocks per q-block row
  constexpr int kQStride = kDB * 128 + 32;  // q/do image: bytes per 4-q row
  constexpr int kXkvB = kKv / 16;           // x image: kv blocks per row
  constexpr int kXStride = kXkvB * 128 + 32;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* q_img = smem;                                  // (kQBlk/4)*kQStride
  char* do_img = q_img + (kQBlk / 4) * kQStride;
  char* x1_img = do_img + (kQBlk / 4) * kQStride;      // P
  char* x2_img = x1_img + (kQBlk / 4) * kXStride;      // dS

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;   // 4 waves x 16 kv rows = 64 kv/block

  int b, h, kv_tile, q_lo, Sq_loc, kv_lo, Skv_loc, causal_off;
  if (cu_q != nullptr) {
    h = blockIdx.y;
    int lo = 0, hi = nseq - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (kvtile_pref[mid] <= (int)blockIdx.x) lo = mid; else hi = mid - 1;
    }
    b = 0;
    kv_tile = blockIdx.x - kvtile_pref[lo];
    q_lo = cu_q[lo];
    Sq_loc = cu_q[lo + 1] - q_lo;
    kv_lo = cu_k[lo];
    Skv_loc = cu_k[lo + 1] - kv_lo;
    causal_off = Skv_loc - Sq_loc + q_offset;
    if (Skv_loc <= 0 || Sq_loc <= 0 || kv_tile * kKv >= Skv_loc) return;
  } else {
    const int nwg = gridDim.x * gridDim.y;
    const int lin = blockIdx.y * gridDim.x + blockIdx.x;
    const int xcd = lin & 7;
    const int qd = nwg >> 3, rd = nwg & 7;
    const int wid = (xcd < rd ? xcd * (qd + 1) : rd * (qd + 1) + (xcd - rd) * qd)
                    + (lin >> 3);
    const int bh = wid / gridDim.x;
    b = bh / Hq;
    h = bh % Hq;
    kv_tile = wid % gridDim.x;
    q_lo = 0; Sq_loc = Sq; kv_lo = 0; Skv_loc = Skv;
    causal_off = q_offset;
  }
  const int hkv = h / (Hq / Hkv);

  const int64_t q_base = ((int64_t)b * Sq * Hq + h) * D;
  const int64_t kv_base = ((int64_t)b * Skv * Hkv + hkv) * D;
  const int64_t q_row_stride = (int64_t)Hq * D;
  const int64_t kv_row_stride = (int64_t)Hkv * D;
  const float* lse_row = lse + ((int64_t)b * Hq + h) * Sq + q_lo;
  const float* delta_row = delta + ((int64_t)b * Hq + h) * Sq + q_lo;

  const int kv0 = kv_tile * kKv;

  // This wave's 16 kv rows. K/V A-fragments are NOT register-resident in
  // v2 (the 64 VGPRs go to occupancy instead): the 16-KB K/V tiles are
  // L2-hot, and each fragment re-load is one coalesced 16-B lane read.
  const int kv_row_global =
      kv_lo + min(kv0 + wave * 16 + (lane & 15), Skv_loc - 1);
  const bf16_t* kp_row = k + kv_base + (int64_t)kv_row_global * kv_row_stride;
  const bf16_t* vp_row = v + kv_base + (int64_t)kv_row_global * kv_row_stride;

  f32x4 dk_acc[kNT], dv_acc[kNT];
#pragma unroll
  for (int nt = 0; nt < kNT; ++nt) {
    dk_acc[nt] = {0.f, 0.f, 0.f, 0.f};
    dv_acc[nt] = {0.f, 0.f, 0.f, 0.f};
  }

  int q_start = 0;
  if (causal) {
    q_start = (max(kv0 - causal_off, 0) / kQBlk) * kQBlk;
  }
  if (q_start >= Sq_loc) return;

  // Q/dO staging straight into the blocked images: thread chunk = one row's
  // 8 consecutive d (16 B), landing contiguously inside a [4 q][16 d] block.
  auto store_qdo = [&](int qt) {
#pragma unroll
    for (int it = 0; it < (kQBlk * D) / (256 * 8); ++it) {
      const int idx = (threadIdx.x + it * 256) * 8;
      const int row = idx / D;
      const int col = idx % D;
      const int g_row = q_lo + min(qt + row, Sq_loc - 1);
      const size_t byte = (size_t)(row >> 2) * kQStride + (col >> 4) * 128 +
                          (row & 3) * 32 + (col & 15) * 2;
      *reinterpret_cast<bf16x8*>(q_img + byte) =
          *reinterpret_cast<const bf16x8*>(
              q + q_base + (int64_t)g_row * q_row_stride + col);
      *reinterpret_cast<bf16x8*>(do_img + byte) =
          *reinterpret_cast<const bf16x8*>(
              dout + q_base + (int64_t)g_row * q_row_stride + col);
    }
  };

  // B fragment (k = 8 d at fixed q row): one 16-B row read of the image.
  auto row_frag = [&](const char* img, int q_row, int d0) -> bf16x8 {
    return *reinterpret_cast<const bf16x8*>(
        img + (size_t)(q_row >> 2) * kQStride + (d0 >> 4) * 128 +
        (q_row & 3) * 32 + (d0 & 15) * 2);
  };
  // B fragment (k = 8 q at fixed d col): tr16 pair of [4 q][16 d] blocks.
  auto tr_frag = [&](const char* img, int q0, int dt) -> bf16x8 {
    const size_t b0 = (size_t)(q0 >> 2) * kQStride + dt * 128 + (lane & 15) * 8;
    bf16x4 lo4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4*)(img + b0));
    bf16x4 hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4*)(img + b0 + kQStride));
    bf16x8 f;
#pragma unroll
    for (int t2 = 0; t2 < 4; ++t2) { f[t2] = lo4[t2]; f[4 + t2] = hi4[t2]; }
    return f;
  };

  store_qdo(q_start);
  __syncthreads();

  for (int qt = q_start; qt < Sq_loc; qt += kQBlk) {
    // ---- S^T, dP^T, softmax — fused PER n-tile so only ONE st/dpt f32x4
    // pair is live at a time (the [4]-array version costs 24 more VGPRs and
    // pushes the kernel past the 168-reg / 3-wave budget). K/V fragment
    // loads repeat per nt; the 16-KB tiles are L2-hot.
    const bool any_mask =
        (kv0 + kKv > Skv_loc) || (qt + kQBlk > Sq_loc) ||
        (causal && kv0 + kKv - 1 > qt + causal_off) ||
        (window_left >= 0 && kv0 < qt + kQBlk - 1 + causal_off - window_left);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int q_row = nt * 16 + (lane & 15);
      f32x4 st_nt = {0.f, 0.f, 0.f, 0.f};
      f32x4 dpt_nt = {0.f, 0.f, 0.f, 0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < kKS; ++ks) {
        const int d0 = ks * 32 + (lane >> 4) * 8;
        const bf16x8 kf = *reinterpret_cast<const bf16x8*>(kp_row + d0);
        st_nt = mfma16(kf, row_frag(q_img, q_row, d0), st_nt);
        const bf16x8 vf = *reinterpret_cast<const bf16x8*>(vp_row + d0);
        dpt_nt = mfma16(vf, row_frag(do_img, q_row, d0), dpt_nt);
      }
      __builtin_amdgcn_s_setprio(0);
      const int q_g = qt + nt * 16 + (lane & 15);
      const float l = (q_g < Sq_loc) ? lse_row[min(q_g, Sq_loc - 1)] : 1e30f;
      const float dlt = (q_g < Sq_loc) ? delta_row[min(q_g, Sq_loc - 1)] : 0.f;
      float pt4[4], dst4[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kv_g = kv0 + wave * 16 + (lane >> 4) * 4 + r;
        float s = st_nt[r] * scale;
        float p;
        if (any_mask) {
          bool masked = (kv_g >= Skv_loc) || (q_g >= Sq_loc);
          if (causal) masked |= kv_g > q_g + causal_off;
          if (window_left >= 0) masked |= kv_g < q_g + causal_off - window_left;
          p = masked ? 0.f : __builtin_amdgcn_exp2f((s - l) * kLog2e);
        } else {
          p = __builtin_amdgcn_exp2f((s - l) * kLog2e);
        }
        pt4[r] = p;
        dst4[r] = p * (dpt_nt[r] - dlt) * scale;
      }
      const int q_l = nt * 16 + (lane & 15);
      const int kv_l = wave * 16 + (lane >> 4) * 4;
      const size_t byte = (size_t)(q_l >> 2) * kXStride + (kv_l >> 4) * 128 +
                          (q_l & 3) * 32 + (kv_l & 15) * 2;
      ushort2_t p4, d4;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p4.s[r] = bf16_bits((bf16_t)pt4[r]);
        d4.s[r] = bf16_bits((bf16_t)dst4[r]);
      }
      *reinterpret_cast<uint64_t*>(x1_img + byte) = p4.u;
      *reinterpret_cast<uint64_t*>(x2_img + byte) = d4.u;
    }
    __syncthreads();

    // ---- dV += P^T @ dO; dK += dS^T @ Q ----------------------------------
    {
      auto xtr = [&](const char* img, int q0) -> bf16x8 {
        const size_t b0 =
            (size_t)(q0 >> 2) * kXStride + wave * 128 + (lane & 15) * 8;
        bf16x4 lo4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (__attribute__((address_space(3))) bf16x4*)(img + b0));
        bf16x4 hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (__attribute__((address_space(3))) bf16x4*)(img + b0 + kXStride));
        bf16x8 f;
#pragma unroll
        for (int t2 = 0; t2 < 4; ++t2) { f[t2] = lo4[t2]; f[4 + t2] = hi4[t2]; }
        return f;
      };
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks2 = 0; ks2 < 2; ++ks2) {
        const int q_off = ks2 * 32 + (lane >> 4) * 8;
        const bf16x8 pa = xtr(x1_img, q_off);
        const bf16x8 da = xtr(x2_img, q_off);
#pragma unroll
        for (int nt = 0; nt < kNT; ++nt) {
          const bf16x8 dob = tr_frag(do_img, q_off, nt);
          const bf16x8 qb = tr_frag(q_img, q_off, nt);
          dv_acc[nt] = mfma16(pa, dob, dv_acc[nt]);
          dk_acc[nt] = mfma16(da, qb, dk_acc[nt]);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }

    // ---- dQ partial = dS @ K ---------------------------------------------
    // do_img is dead once the dV mfma above has consumed it: re-stage the
    // 16-KB K tile THERE as the same [4 kv][16 d] blocked image (tr_frag
    // pairs give the B[k=kv][n=d] fragments). A global-gather alternative
    // (8 ushort loads + perms per fragment) measured 38 scratch spill ops:
    // the hoisted gathers blew the 168-reg budget.
    __syncthreads();
    {
#pragma unroll
      for (int it = 0; it < (kKv * D) / (256 * 8); ++it) {
        const int idx = (threadIdx.x + it * 256) * 8;
        const int row = idx / D;
        const int col = idx % D;
        const int g_row = kv_lo + min(kv0 + row, Skv_loc - 1);
        const size_t byte = (size_t)(row >> 2) * kQStride + (col >> 4) * 128 +
                            (row & 3) * 32 + (col & 15) * 2;
        *reinterpret_cast<bf16x8*>(do_img + byte) =
            *reinterpret_cast<const bf16x8*>(
                k + kv_base + (int64_t)g_row * kv_row_stride + col);
      }
    }
    __syncthreads();
    {
      // 4 waves x (q m-tile = wave); d handled in full by each wave
      const int q_mt = wave;
      const int q_row = q_mt * 16 + (lane & 15);
      const size_t qbase = (size_t)(q_row >> 2) * kXStride + (q_row & 3) * 32;
      f32x4 dq_acc[kNT];
#pragma unroll
      for (int nt = 0; nt < kNT; ++nt) dq_acc[nt] = {0.f, 0.f, 0.f, 0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks2 = 0; ks2 < 2; ++ks2) {
        const int kv_off = ks2 * 32 + (lane >> 4) * 8;
        const bf16x8 da = *reinterpret_cast<const bf16x8*>(
            x2_img + qbase + (kv_off >> 4) * 128 + (kv_off & 15) * 2);
#pragma unroll
        for (int nt = 0; nt < kNT; ++nt) {
          dq_acc[nt] = mfma16(da, tr_frag(do_img, kv_off, nt), dq_acc[nt]);
        }
      }
      __builtin_amdgcn_s_setprio(0);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int q_g = qt + q_mt * 16 + (lane >> 4) * 4 + r;
        if (q_g < Sq_loc) {
#pragma unroll
          for (int nt = 0; nt < kNT; ++nt) {
            atomicAdd(dq + q_base +
                          (int64_t)(q_lo + q_g) * q_row_stride + nt * 16 +
                          (lane & 15),
                      dq_acc[nt][r]);
          }
        }
      }
    }
    // next tile's Q/dO overwrite the images: wait for all readers
    __syncthreads();
    if (qt + kQBlk < Sq_loc) {
      store_qdo(qt + kQBlk);
      __syncthreads();
    }
  }

  // ---- flush dK, dV (atomicAdd: GQA groups and padded tiles overlap) -------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kv_g = kv0 + wave * 16 + (lane >> 4) * 4 + r;
    if (kv_g < Skv_loc) {
#pragma unroll
      for (int nt = 0; nt < kNT; ++nt) {
        const int d = nt * 16 + (lane & 15);
        atomicAdd(dk + kv_base + (int64_t)(kv_lo + kv_g) * kv_row_stride + d,
                  dk_acc[nt][r]);
        atomicAdd(dv + kv_base + (int64_t)(kv_lo + kv_g) * kv_row_stride + d,
                  dv_acc[nt][r]);
      }
    }
  }
}

// MFMA fragment-map self-check: C = A@B for one 16x32 @ 32x16 tile.
__global__ void mfma_selfcheck_kernel(
    const bf16_t* __restrict__ a,  // (16, 32) row-major
    const bf16_t* __restrict__ b,  // (32, 16) row-major
    float* __restrict__ c) {       // (16, 16) row-major
  const int lane = threadIdx.x & 63;
  bf16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = a[(lane & 15) * 32 + (lane >> 4) * 8 + j];
    bf[j] = b[((lane >> 4) * 8 + j) * 16 + (lane & 15)];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = mfma16(af, bf, acc);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    c[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
  }
}

// Fused fp32 -> bf16 cast of the three backward accumulators in ONE launch
// (the three at::native casts measured ~2.6 TB/s and ~0.7 ms per backward at
// bench shape). Sizes are multiples of 8 (B*S*H*D), so 8-float groups never
// straddle a buffer boundary.
__global__ void cast3_f32_bf16_kernel(
    const float* __restrict__ a, const float* __restrict__ b,
    const float* __restrict__ c,
    ushort* __restrict__ oa, ushort* __restrict__ ob, ushort* __restrict__ oc,
    int64_t na, int64_t nb, int64_t nc) {
  const int64_t total = na + nb + nc;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       i < total; i += stride) {
    const float* src;
    ushort* dst;
    int64_t off = i;
    if (off < na) {
      src = a; dst = oa;
    } else if (off < na + nb) {
      src = b; dst = ob; off -= na;
    } else {
      src = c; dst = oc; off -= na + nb;
    }
    f32x4 lo = *reinterpret_cast<const f32x4*>(src + off);
    f32x4 hi = *reinterpret_cast<const f32x4*>(src + off + 4);
    ushort2_t p0, p1;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      p0.s[j] = bf16_bits((bf16_t)lo[j]);
      p1.s[j] = bf16_bits((bf16_t)hi[j]);
    }
    *reinterpret_cast<uint64_t*>(dst + off) = p0.u;
    *reinterpret_cast<uint64_t*>(dst + off + 4) = p1.u;
  }
}

}  // namespace d9d

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

namespace {

int padded_head_dim(int d) {
  for (int cand : {32, 64, 96, 128}) {
    if (d <= cand) return cand;
  }
  TORCH_CHECK(false, "head_dim too large: ", d);
  return -1;
}

torch::Tensor maybe_pad_d(torch::Tensor t, int D_pad) {
  const int D = t.size(-1);
  if (D == D_pad) return t.contiguous();
  auto padded = torch::zeros(
      {t.size(0), t.size(1), t.size(2), (int64_t)D_pad}, t.options());
  padded.narrow(-1, 0, D).copy_(t);
  return padded;
}

// Per-sequence tile-count prefix for varlen grids (CPU int32 cu_seqlens).
std::pair<torch::Tensor, int> tile_prefix(torch::Tensor cu_cpu, int tile) {
  const int n = cu_cpu.numel() - 1;
  auto pref = torch::empty({n + 1}, torch::dtype(torch::kInt32));
  const int* cu = cu_cpu.data_ptr<int>();
  int* p = pref.data_ptr<int>();
  int total = 0;
  for (int i = 0; i < n; ++i) {
    p[i] = total;
    total += (cu[i + 1] - cu[i] + tile - 1) / tile;
  }
  p[n] = total;
  return {pref, total};
}

}  // namespace

std::vector<torch::Tensor> flash_attn_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    c10::optional<torch::Tensor> sinks,
    c10::optional<torch::Tensor> cu_seqlens_q,
    c10::optional<torch::Tensor> cu_seqlens_k,
    bool causal, double softmax_scale, int64_t window_left, int64_t q_offset) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  const bool varlen = cu_seqlens_q.has_value();
  if (varlen) {
    TORCH_CHECK(q.dim() == 3, "varlen expects (total, H, D)");
    q = q.unsqueeze(0);
    k = k.unsqueeze(0);
    v = v.unsqueeze(0);
  }
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4);
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Skv = k.size(1), Hkv = k.size(2);
  TORCH_CHECK(Hq % Hkv == 0, "GQA head mismatch");

  const int D_pad = padded_head_dim(D);
  auto qp = maybe_pad_d(q, D_pad);
  auto kp = maybe_pad_d(k, D_pad);
  auto vp = maybe_pad_d(v, D_pad);

  torch::Tensor cu_q_gpu, cu_k_gpu, qpref_gpu;
  const int* cu_q_ptr = nullptr;
  const int* cu_k_ptr = nullptr;
  const int* qpref_ptr = nullptr;
  int nseq = 0, n_qtiles = 0;
  if (varlen) {
    auto cu_q_cpu = cu_seqlens_q->to(torch::kInt32).cpu().contiguous();
    auto cu_k_cpu = cu_seqlens_k->to(torch::kInt32).cpu().contiguous();
    nseq = cu_q_cpu.numel() - 1;
    auto [pref, total] = tile_prefix(cu_q_cpu, 128);
    n_qtiles = total;
    cu_q_gpu = cu_q_cpu.to(q.device(), true);
    cu_k_gpu = cu_k_cpu.to(q.device(), true);
    qpref_gpu = pref.to(q.device(), true);
    cu_q_ptr = cu_q_gpu.data_ptr<int>();
    cu_k_ptr = cu_k_gpu.data_ptr<int>();
    qpref_ptr = qpref_gpu.data_ptr<int>();
  }

  auto out = torch::empty_like(qp);
  auto lse = torch::empty({varlen ? 1 : B, Hq, Sq},
                          q.options().dtype(torch::kFloat32));
  const float* sinks_ptr = nullptr;
  torch::Tensor sinks_f;
  if (sinks.has_value()) {
    sinks_f = sinks->to(torch::kFloat32).contiguous();
    TORCH_CHECK(sinks_f.numel() == Hq, "sinks must be (Hq,)");
    sinks_ptr = sinks_f.data_ptr<float>();
  }

  const dim3 grid(varlen ? n_qtiles : (Sq + 127) / 128,
                  varlen ? Hq : B * Hq);
  const size_t smem =
      (size_t)(d9d::kKvBlk * D_pad + D_pad * d9d::kKvBlk + 128 * (d9d::kKvBlk + 8)) *
      sizeof(__bf16);
  auto stream = at::hip::getCurrentHIPStream();

#define LAUNCH_FWD(DP)                                                        \
  hipLaunchKernelGGL((d9d::flash_fwd_kernel<DP>), grid, dim3(256), smem,      \
                     stream,                                                  \
                     reinterpret_cast<const __bf16*>(qp.data_ptr()),          \
                     reinterpret_cast<const __bf16*>(kp.data_ptr()),          \
                     reinterpret_cast<const __bf16*>(vp.data_ptr()),          \
                     reinterpret_cast<__bf16*>(out.data_ptr()),               \
                     lse.data_ptr<float>(), sinks_ptr, cu_q_ptr, cu_k_ptr,    \
                     qpref_ptr, nseq, B, Sq, Skv, Hq, Hkv,                    \
                     (float)softmax_scale, causal ? 1 : 0, (int)window_left,   \
                     (int)q_offset)
  switch (D_pad) {
    case 32: LAUNCH_FWD(32); break;
    case 64: LAUNCH_FWD(64); break;
    case 96: LAUNCH_FWD(96); break;
    case 128: LAUNCH_FWD(128); break;
  }
#undef LAUNCH_FWD

  if (D_pad != D) out = out.narrow(-1, 0, D).contiguous();
  if (varlen) {
    out = out.squeeze(0);
    lse = lse.squeeze(0);  // (Hq, total)
  }
  return {out, lse};
}

std::vector<torch::Tensor> flash_attn_bwd(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor lse,
    c10::optional<torch::Tensor> cu_seqlens_q,
    c10::optional<torch::Tensor> cu_seqlens_k,
    bool causal, double softmax_scale, int64_t window_left, int64_t q_offset) {
  const bool varlen = cu_seqlens_q.has_value();
  if (varlen) {
    dout = dout.unsqueeze(0);
    q = q.unsqueeze(0);
    k = k.unsqueeze(0);
    v = v.unsqueeze(0);
    out = out.unsqueeze(0);
    lse = lse.unsqueeze(0);
  }
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Skv = k.size(1), Hkv = k.size(2);
  const int D_pad = padded_head_dim(D);

  torch::Tensor cu_q_gpu, cu_k_gpu, kvpref_gpu;
  const int* cu_q_ptr = nullptr;
  const int* cu_k_ptr = nullptr;
  const int* kvpref_ptr = nullptr;
  int nseq = 0, n_kvtiles = 0;
  if (varlen) {
    auto cu_q_cpu = cu_seqlens_q->to(torch::kInt32).cpu().contiguous();
    auto cu_k_cpu = cu_seqlens_k->to(torch::kInt32).cpu().contiguous();
    nseq = cu_q_cpu.numel() - 1;
    auto [pref, total] = tile_prefix(cu_k_cpu, d9d::kBwdKv);
    n_kvtiles = total;
    cu_q_gpu = cu_q_cpu.to(q.device(), true);
    cu_k_gpu = cu_k_cpu.to(q.device(), true);
    kvpref_gpu = pref.to(q.device(), true);
    cu_q_ptr = cu_q_gpu.data_ptr<int>();
    cu_k_ptr = cu_k_gpu.data_ptr<int>();
    kvpref_ptr = kvpref_gpu.data_ptr<int>();
  }

  auto qp = maybe_pad_d(q, D_pad);
  auto kp = maybe_pad_d(k, D_pad);
  auto vp = maybe_pad_d(v, D_pad);
  auto dop = maybe_pad_d(dout, D_pad);
  auto op = maybe_pad_d(out, D_pad);

  auto f32opt = q.options().dtype(torch::kFloat32);
  auto delta = torch::empty({B, Hq, Sq}, f32opt);
  // atomicAdd accumulators: cleared via the copy/DMA engine (hipMemsetAsync)
  // instead of torch::zeros — the fill kernels measured ~1.7 TB/s in-step
  // and 400 MB/call of clears showed up at ~1.3% of the whole bench step.
  auto dq32 = torch::empty({B, Sq, Hq, D_pad}, f32opt);
  auto dk32 = torch::empty({B, Skv, Hkv, D_pad}, f32opt);
  auto dv32 = torch::empty({B, Skv, Hkv, D_pad}, f32opt);

  auto stream = at::hip::getCurrentHIPStream();
  hipMemsetAsync(dq32.data_ptr(), 0, dq32.numel() * sizeof(float), stream);
  hipMemsetAsync(dk32.data_ptr(), 0, dk32.numel() * sizeof(float), stream);
  hipMemsetAsync(dv32.data_ptr(), 0, dv32.numel() * sizeof(float), stream);
  {
    const int64_t rows = (int64_t)B * Sq * Hq;
    hipLaunchKernelGGL(d9d::attn_delta_kernel, dim3(rows), dim3(64), 0, stream,
                       reinterpret_cast<const __bf16*>(dop.data_ptr()),
                       reinterpret_cast<const __bf16*>(op.data_ptr()),
                       delta.data_ptr<float>(), B, Sq, Hq, D_pad);
  }

  // v2 geometry experiment (kv-64 tiles, 3 workgroups/CU): opt-in via
  // D9D_FLASH_BWD_V2=1, non-varlen only.
  static const bool use_bwd_v2 = []() {
    const char* e = getenv("D9D_FLASH_BWD_V2");
    return e != nullptr && e[0] == '1';
  }();
  const bool run_v2 = use_bwd_v2 && !varlen;
  if (run_v2) {
    const dim3 grid2((Skv + 63) / 64, B * Hq);
    const size_t qs = (size_t)(d9d::kQBlk / 4) * ((D_pad / 16) * 128 + 32);
    const size_t xs = (size_t)(d9d::kQBlk / 4) * ((64 / 16) * 128 + 32);
    const size_t smem2 = 2 * qs + 2 * xs;
#define LAUNCH_BWD2(DP)                                                       \
  hipLaunchKernelGGL((d9d::flash_bwd_kernel_v2<DP>), grid2, dim3(256), smem2, \
                     stream,                                                  \
                     reinterpret_cast<const __bf16*>(qp.data_ptr()),          \
                     reinterpret_cast<const __bf16*>(kp.data_ptr()),          \
                     reinterpret_cast<const __bf16*>(vp.data_ptr()),          \
                     reinterpret_cast<const __bf16*>(dop.data_ptr()),         \
                     lse.data_ptr<float>(), delta.data_ptr<float>(),          \
                     dq32.data_ptr<float>(), dk32.data_ptr<float>(),          \
                     dv32.data_ptr<float>(), nullptr, nullptr, nullptr, 0,    \
                     B, Sq, Skv, Hq, Hkv, (float)softmax_scale,               \
                     causal ? 1 : 0, (int)window_left, (int)q_offset)
    switch (D_pad) {
      case 32: LAUNCH_BWD2(32); break;
      case 64: LAUNCH_BWD2(64); break;
      case 96: LAUNCH_BWD2(96); break;
      case 128: LAUNCH_BWD2(128); break;
    }
#undef LAUNCH_BWD2
  }

  const dim3 grid(varlen ? n_kvtiles : (Skv + d9d::kBwdKv - 1) / d9d::kBwdKv,
                  varlen ? Hq : B * Hq);
  // 4 q/do/t-images + K^T + the two blocked P/dS images (padded q-block
  // rows: (kBwdKv/16)*128 + 32 bytes per 4 q rows)
  const size_t smem =
      (size_t)(4 * d9d::kQBlk * D_pad + D_pad * d9d::kBwdKv) * sizeof(__bf16)
      + 2 * (size_t)(d9d::kQBlk / 4) * ((d9d::kBwdKv / 16) * 128 + 32);

#define LAUNCH_BWD(DP)                                                        \
  hipLaunchKernelGGL((d9d::flash_bwd_kernel<DP>), grid, dim3(512), smem,      \
                     stream,                                                  \
                     reinterpret_cast<const __bf16*>(qp.data_ptr()),          \
                     reinterpret_cast<const __bf16*>(kp.data_ptr()),          \
                     reinterpret_cast<const __bf16*>(vp.data_ptr()),          \
                     reinterpret_cast<const __bf16*>(dop.data_ptr()),         \
                     lse.data_ptr<float>(), delta.data_ptr<float>(),          \
                     dq32.data_ptr<float>(), dk32.data_ptr<float>(),          \
                     dv32.data_ptr<float>(), cu_q_ptr, cu_k_ptr, kvpref_ptr,  \
                     nseq, B, Sq, Skv, Hq, Hkv,                               \
                     (float)softmax_scale, causal ? 1 : 0, (int)window_left,   \
                     (int)q_offset)
  if (!run_v2) {
    switch (D_pad) {
      case 32: LAUNCH_BWD(32); break;
      case 64: LAUNCH_BWD(64); break;
      case 96: LAUNCH_BWD(96); break;
      case 128: LAUNCH_BWD(128); break;
    }
  }
#undef LAUNCH_BWD

  torch::Tensor dq, dk, dv;
  if (D == D_pad) {
    // one fused cast launch for all three accumulators
    auto bf = q.options().dtype(torch::kBFloat16);
    dq = torch::empty(dq32.sizes(), bf);
    dk = torch::empty(dk32.sizes(), bf);
    dv = torch::empty(dv32.sizes(), bf);
    const int64_t na = dq32.numel(), nb = dk32.numel(), nc = dv32.numel();
    const int64_t groups = (na + nb + nc) / 8;
    const int blocks = (int)std::min<int64_t>((groups + 255) / 256, 16384);
    hipLaunchKernelGGL(d9d::cast3_f32_bf16_kernel, dim3(blocks), dim3(256), 0,
                       stream, dq32.data_ptr<float>(), dk32.data_ptr<float>(),
                       dv32.data_ptr<float>(),
                       reinterpret_cast<ushort*>(dq.data_ptr()),
                       reinterpret_cast<ushort*>(dk.data_ptr()),
                       reinterpret_cast<ushort*>(dv.data_ptr()),
                       na, nb, nc);
  } else {
    dq = dq32.narrow(-1, 0, D).to(torch::kBFloat16).contiguous();
    dk = dk32.narrow(-1, 0, D).to(torch::kBFloat16).contiguous();
    dv = dv32.narrow(-1, 0, D).to(torch::kBFloat16).contiguous();
  }
  if (varlen) {
    dq = dq.squeeze(0);
    dk = dk.squeeze(0);
    dv = dv.squeeze(0);
  }
  return {dq, dk, dv};
}

torch::Tensor mfma_selfcheck(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(b.sizes() == torch::IntArrayRef({32, 16}));
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(d9d::mfma_selfcheck_kernel, dim3(1), dim3(64), 0, stream,
                     reinterpret_cast<const __bf16*>(a.contiguous().data_ptr()),
                     reinterpret_cast<const __bf16*>(b.contiguous().data_ptr()),
                     c.data_ptr<float>());
  return c;
}
