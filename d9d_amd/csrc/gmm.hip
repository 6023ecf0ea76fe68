// CDNA4 grouped GEMM for MoE expert compute.
//
// Replaces the reference's nv-grouped-gemm wheel (d9d/kernel/gmm/function.py).
// out[rows_e] = a[rows_e] @ b[e] for each expert e; row counts are ragged.
// Host builds tiny per-expert offset tables (row_offsets, m-tile prefix) from
// the CPU batch_sizes; ONE kernel launch covers every expert — each block
// binary-searches its m-tile, so there is no per-expert launch overhead
// (the eager fallback was E x 3 rocBLAS launches per MoE layer).
//
// Geometry: 256x256 tiles, 8 waves (2 m-halves x 4 n-quarters), BK=64,
// T14 register-staged LDS with the ((row&7)<<4) XOR swizzle for
// conflict-free ds_read_b128, v_mfma_f32_16x16x32_bf16.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <map>
#include <mutex>
#include <tuple>

namespace d9d {

typedef __bf16 bf16_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;

D9D_DEVICE f32x4 mfma16g(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

constexpr int kBM = 256;
constexpr int kBN = 192;
constexpr int kBK = 64;

// A LDS tile: [kBM][kBK] row-major, rows kBK*2=128 bytes -> swizzle mask 7.
// B LDS tile staged TRANSPOSED as bt[n][k] (kBN x kBK): the MFMA B-fragment
// needs 8 contiguous k at fixed n; b (E,K,N) is n-contiguous, so staging
// reads 8 consecutive n at fixed k and scatters 8 scalar LDS writes.
//
// 256x192 tiles with 8 waves (2 m-halves x 4 n-quarters of 48): this kernel is
// HBM-bound on A re-reads (once per n-tile), and the big tile quarters that
// traffic vs the old 128x64 geometry. Register (T14) staging doubles as the
// second buffer: loads for k-tile t+1 issue early under tile t's MFMAs.

__global__ __launch_bounds__(512, 1) void gmm_kernel(
    const bf16_t* __restrict__ a,    // (T, K)
    const bf16_t* __restrict__ b,    // (E, K, N)
    bf16_t* __restrict__ out,        // (T, N)
    const int* __restrict__ row_off,      // (E+1,)
    const int* __restrict__ mtile_pref,   // (E+1,) prefix of ceil(rows_e/kBM)
    int E, int K, int N, int n_tiles) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* a_lds = reinterpret_cast<bf16_t*>(smem);        // [kBM][kBK]
  bf16_t* bt_lds = a_lds + kBM * kBK;                     // [kBN][kBK]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;   // 0..1: wave row (128 rows each)
  const int wn = wave & 3;    // 0..3: wave col (64 cols each)

  // ---- XCD-aware expert-major work mapping (see gmm_nt_kernel) -------------
  const int nwg = gridDim.x;
  const int xcd = blockIdx.x & 7;
  const int q = nwg >> 3, r = nwg & 7;
  const int wid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                  + (blockIdx.x >> 3);
  // ---- map work id -> (expert, m_tile) via binary search -------------------
  const int mt_global = wid / n_tiles;
  int lo = 0, hi = E - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (mtile_pref[mid] <= mt_global) lo = mid; else hi = mid - 1;
  }
  const int e = lo;
  const int m_tile = mt_global - mtile_pref[e];
  const int row0 = row_off[e] + m_tile * kBM;
  const int row_end = row_off[e + 1];
  const int n0 = (wid % n_tiles) * kBN;

  const bf16_t* b_e = b + (int64_t)e * K * N;

  f32x4 acc[8][3];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int n_ktiles = (K + kBK - 1) / kBK;

  // T14 split staging: per k-tile, 4x bf16x8 of A and 4x bf16x8 of B per
  // thread are loaded into registers EARLY (issue overlaps the previous
  // tile's MFMAs) and written to LDS late.
  union bvec {
    bf16x8 v;
    ushort s[8];
  };
  bf16x8 a_reg[4];
  bvec b_reg[2][2];

  // Branchless staging: clamp addresses into range, select-zero after the
  // load (a per-element branch around a load makes hipcc serialize every
  // load behind a vmcnt(0) drain).
  auto load_regs = [&](int kt) {
    const int k0 = kt * kBK;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = (threadIdx.x + it * 512) * 8;
      const int row = idx / kBK;
      const int col = idx % kBK;
      const int g_row = row0 + row;
      const bool ok = g_row < row_end && k0 + col + 7 < K;
      const int64_t sr = min(g_row, row_end - 1);
      const int64_t sc = min(k0 + col, max(K - 8, 0));
      const bf16x8 val = *reinterpret_cast<const bf16x8*>(a + sr * K + sc);
      if (ok) {
        a_reg[it] = val;
      } else {
        // K-tail / row-tail edge (rare): per-element fill
        bf16x8 ev = {};
        if (g_row < row_end) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            if (k0 + col + j < K) ev[j] = a[(int64_t)g_row * K + k0 + col + j];
          }
        }
        a_reg[it] = ev;
      }
    }
    // B: (kBK/2) x (kBN/8) = 768 k-pair groups; each group loads two
    // bf16x8 (adjacent k, same 8 n) so the LDS writes pair into u32s.
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int grp = threadIdx.x + it * 512;
      if (grp >= (kBK / 2) * (kBN / 8)) break;
      const int kk = (grp / (kBN / 8)) * 2;
      const int n = (grp % (kBN / 8)) * 8;
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const bool ok = k0 + kk + h < K && n0 + n + 7 < N;
        const int64_t sk = min(k0 + kk + h, K - 1);
        const int64_t sn = min(n0 + n, max(N - 8, 0));
        const bf16x8 val = *reinterpret_cast<const bf16x8*>(b_e + sk * N + sn);
        if (ok) {
          b_reg[it][h].v = val;
        } else {
          // slow edge path (rare): per-element fill
          bf16x8 ev = {};
          if (k0 + kk + h < K) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              if (n0 + n + j < N) ev[j] = b_e[(int64_t)(k0 + kk + h) * N + n0 + n + j];
            }
          }
          b_reg[it][h].v = ev;
        }
      }
    }
  };

  auto store_lds = [&]() {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = (threadIdx.x + it * 512) * 8;
      const int row = idx / kBK;
      const int col = idx % kBK;
      const int byte = (col * 2) ^ ((row & 7) << 4);
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(a_lds) + row * (kBK * 2) + byte) = a_reg[it];
    }
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int grp = threadIdx.x + it * 512;
      if (grp >= (kBK / 2) * (kBN / 8)) break;
      const int kk = (grp / (kBN / 8)) * 2;
      const int n = (grp % (kBN / 8)) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int nn = n + j;
        const uint32_t u = (uint32_t)b_reg[it][0].s[j] |
                           ((uint32_t)b_reg[it][1].s[j] << 16);
        // swizzle on (nn ^ nn>>3): the paired writes walk nn in +8 steps
        // inside one instruction (nn&7 alone is constant there -> 24-way
        // bank conflict, PMC-measured 21% of wave cycles); the reads walk
        // nn in +1 steps, and the combined mask spreads both 8 ways.
        const int byte = (kk * 2) ^ ((((nn >> 3) ^ nn) & 7) << 4);
        *reinterpret_cast<uint32_t*>(
            reinterpret_cast<char*>(bt_lds) + nn * (kBK * 2) + byte) = u;
      }
    }
  };

  load_regs(0);
  store_lds();
  __syncthreads();

  for (int kt = 0; kt < n_ktiles; ++kt) {
    if (kt + 1 < n_ktiles) load_regs(kt + 1);  // issue early, use late

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // kBK=64 -> 2 MFMA k-steps
      const int kk = ks * 32 + (lane >> 4) * 8;
      bf16x8 a_frag[8], b_frag[3];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int row = wm * 128 + i * 16 + (lane & 15);
        const int byte = (kk * 2) ^ ((row & 7) << 4);
        a_frag[i] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(a_lds) + row * (kBK * 2) + byte);
      }
#pragma unroll
      for (int j = 0; j < 3; ++j) {
        const int col = wn * 48 + j * 16 + (lane & 15);
        const int byte = (kk * 2) ^ ((((col >> 3) ^ col) & 7) << 4);
        b_frag[j] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(bt_lds) + col * (kBK * 2) + byte);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 3; ++j) acc[i][j] = mfma16g(a_frag[i], b_frag[j], acc[i][j]);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    if (kt + 1 < n_ktiles) {
      store_lds();
      __syncthreads();
    }
  }

  // ---- store ---------------------------------------------------------------
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 3; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = row0 + wm * 128 + i * 16 + (lane >> 4) * 4 + r;
        const int col = n0 + wn * 48 + j * 16 + (lane & 15);
        if (row < row_end && col < N) {
          out[(int64_t)row * N + col] = (bf16_t)acc[i][j][r];
        }
      }
    }
  }
}

// out[rows_e] = a[rows_e] @ w[e]^T for w stored (E, N, K) (nn.Linear layout).
// Both operands are k-contiguous, so BOTH stage as plain vector copies into
// row-major swizzled LDS tiles -- no scatter. Same 256x192 / 8-wave geometry
// as gmm_kernel; this is the fast path for expert FORWARD. The dgrad
// (dy @ w with w (N,K) read as (red,out)) reuses gmm_kernel unchanged.
__global__ __launch_bounds__(512, 1) void gmm_nt_kernel(
    const bf16_t* __restrict__ a,    // (T, K)
    const bf16_t* __restrict__ w,    // (E, N, K)
    bf16_t* __restrict__ out,        // (T, N)
    const int* __restrict__ row_off,
    const int* __restrict__ mtile_pref,
    int E, int K, int N, int n_tiles) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* a_lds = reinterpret_cast<bf16_t*>(smem);        // [kBM][kBK]
  bf16_t* bt_lds = a_lds + kBM * kBK;                     // [kBN][kBK]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;

  // XCD-aware work mapping: hardware deals linear block ids round-robin over
  // the 8 XCDs (each with a private L2). The bijective remap gives each XCD a
  // CONTIGUOUS span of expert-major work items, so one expert's weight slice
  // and A rows are re-read from that XCD's L2 instead of HBM.
  const int nwg = gridDim.x;
  const int xcd = blockIdx.x & 7;
  const int q = nwg >> 3, r = nwg & 7;
  const int wid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                  + (blockIdx.x >> 3);
  const int mt_global = wid / n_tiles;
  int lo = 0, hi = E - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (mtile_pref[mid] <= mt_global) lo = mid; else hi = mid - 1;
  }
  const int e = lo;
  const int m_tile = mt_global - mtile_pref[e];
  const int row0 = row_off[e] + m_tile * kBM;
  const int row_end = row_off[e + 1];
  const int n0 = (wid % n_tiles) * kBN;

  const bf16_t* w_e = w + (int64_t)e * N * K;

  f32x4 acc[8][3];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int n_ktiles = (K + kBK - 1) / kBK;

  bf16x8 a_reg[4], b_reg[3];

  auto load_regs = [&](int kt) {
    const int k0 = kt * kBK;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = (threadIdx.x + it * 512) * 8;
      const int row = idx / kBK;
      const int col = idx % kBK;
      const int g_row = row0 + row;
      const bool ok = g_row < row_end && k0 + col + 7 < K;
      const int64_t sr = min(g_row, row_end - 1);
      const int64_t sc = min(k0 + col, max(K - 8, 0));
      const bf16x8 val = *reinterpret_cast<const bf16x8*>(a + sr * K + sc);
      if (ok) {
        a_reg[it] = val;
      } else {
        bf16x8 ev = {};
        if (g_row < row_end) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            if (k0 + col + j < K) ev[j] = a[(int64_t)g_row * K + k0 + col + j];
          }
        }
        a_reg[it] = ev;
      }
    }
#pragma unroll
    for (int it = 0; it < 3; ++it) {
      const int idx = (threadIdx.x + it * 512) * 8;
      const int n = idx / kBK;
      const int col = idx % kBK;
      const bool ok = n0 + n < N && k0 + col + 7 < K;
      const int64_t sn = min(n0 + n, N - 1);
      const int64_t sc = min(k0 + col, max(K - 8, 0));
      const bf16x8 val = *reinterpret_cast<const bf16x8*>(w_e + sn * K + sc);
      if (ok) {
        b_reg[it] = val;
      } else {
        bf16x8 ev = {};
        if (n0 + n < N) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            if (k0 + col + j < K) ev[j] = w_e[(int64_t)(n0 + n) * K + k0 + col + j];
          }
        }
        b_reg[it] = ev;
      }
    }
  };

  auto store_lds = [&]() {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = (threadIdx.x + it * 512) * 8;
      const int row = idx / kBK;
      const int col = idx % kBK;
      const int byte = (col * 2) ^ ((row & 7) << 4);
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(a_lds) + row * (kBK * 2) + byte) = a_reg[it];
    }
#pragma unroll
    for (int it = 0; it < 3; ++it) {
      const int idx = (threadIdx.x + it * 512) * 8;
      const int n = idx / kBK;
      const int col = idx % kBK;
      const int byte = (col * 2) ^ ((n & 7) << 4);
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(bt_lds) + n * (kBK * 2) + byte) = b_reg[it];
    }
  };

  load_regs(0);
  store_lds();
  __syncthreads();

  for (int kt = 0; kt < n_ktiles; ++kt) {
    if (kt + 1 < n_ktiles) load_regs(kt + 1);

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int kk = ks * 32 + (lane >> 4) * 8;
      bf16x8 a_frag[8], b_frag[3];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int row = wm * 128 + i * 16 + (lane & 15);
        const int byte = (kk * 2) ^ ((row & 7) << 4);
        a_frag[i] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(a_lds) + row * (kBK * 2) + byte);
      }
#pragma unroll
      for (int j = 0; j < 3; ++j) {
        const int col = wn * 48 + j * 16 + (lane & 15);
        const int byte = (kk * 2) ^ ((col & 7) << 4);
        b_frag[j] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(bt_lds) + col * (kBK * 2) + byte);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 3; ++j) acc[i][j] = mfma16g(a_frag[i], b_frag[j], acc[i][j]);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    if (kt + 1 < n_ktiles) {
      store_lds();
      __syncthreads();
    }
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 3; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = row0 + wm * 128 + i * 16 + (lane >> 4) * 4 + r;
        const int col = n0 + wn * 48 + j * 16 + (lane & 15);
        if (row < row_end && col < N) {
          out[(int64_t)row * N + col] = (bf16_t)acc[i][j][r];
        }
      }
    }
  }
}

// glds variant of gmm_nt: async global->LDS staging
// (global_load_lds_dwordx4) into a double buffer -- no staging registers, no
// ds_write pass, loads are fire-and-forget behind a counted s_waitcnt
// vmcnt(N). Raw s_barrier (NOT __syncthreads) keeps hipcc from draining the
// in-flight glds at each barrier. The swizzled LDS image is produced by
// PRE-SWIZZLING the per-lane global source address (glds writes linearly:
// wave-uniform base + lane*16). Requires K % 64 == 0 (no k-tail masking is
// possible with direct-to-LDS loads); the register-staged kernel remains the
// fallback.
__global__ __launch_bounds__(512, 1) void gmm_nt_glds_kernel(
    const bf16_t* __restrict__ a,    // (T, K)
    const bf16_t* __restrict__ w,    // (E, N, K)
    bf16_t* __restrict__ out,        // (T, N)
    const int* __restrict__ row_off,
    const int* __restrict__ mtile_pref,
    int E, int K, int N, int n_tiles) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // layout: A0 (32 KB) | B0 (24 KB) | A1 | B1
  char* a_buf[2];
  char* b_buf[2];
  a_buf[0] = smem;
  b_buf[0] = smem + kBM * kBK * 2;
  a_buf[1] = b_buf[0] + kBN * kBK * 2;
  b_buf[1] = a_buf[1] + kBM * kBK * 2;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;

  const int nwg = gridDim.x;
  const int xcd = blockIdx.x & 7;
  const int q = nwg >> 3, r = nwg & 7;
  const int wid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                  + (blockIdx.x >> 3);
  const int mt_global = wid / n_tiles;
  int lo = 0, hi = E - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (mtile_pref[mid] <= mt_global) lo = mid; else hi = mid - 1;
  }
  const int e = lo;
  const int m_tile = mt_global - mtile_pref[e];
  const int row0 = row_off[e] + m_tile * kBM;
  const int row_end = row_off[e + 1];
  const int n0 = (wid % n_tiles) * kBN;

  const bf16_t* w_e = w + (int64_t)e * N * K;

  f32x4 acc[8][3];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int n_ktiles = K / kBK;

  // per k-tile: A = 32 wave-chunks of 1 KB, B = 24; wave 'wave' stages
  // A chunks {wave + 8i} and B chunks {wave + 8i}.
  auto stage = [&](int kt, int buf) {
    const int k0 = kt * kBK;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int chunk = wave + i * 8;
      const int l = chunk * 1024 + lane * 16;   // linear LDS byte
      const int arow = l >> 7;                  // 128-byte rows
      const int colb = (l & 127) ^ ((arow & 7) << 4);  // source pre-swizzle
      const int64_t sr = min(row0 + arow, row_end - 1);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const uint32_t*>(
              reinterpret_cast<const char*>(a) + (sr * (int64_t)K + k0) * 2 + colb),
          reinterpret_cast<uint32_t*>(a_buf[buf] + chunk * 1024), 16, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < 3; ++i) {
      const int chunk = wave + i * 8;
      const int l = chunk * 1024 + lane * 16;
      const int nrow = l >> 7;
      const int colb = (l & 127) ^ ((nrow & 7) << 4);
      const int64_t sn = min(n0 + nrow, N - 1);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const uint32_t*>(
              reinterpret_cast<const char*>(w_e) + (sn * (int64_t)K + k0) * 2 + colb),
          reinterpret_cast<uint32_t*>(b_buf[buf] + chunk * 1024), 16, 0, 0);
    }
  };

  stage(0, 0);
  if (n_ktiles > 1) stage(1, 1);

  for (int kt = 0; kt < n_ktiles; ++kt) {
    const int buf = kt & 1;
    // drain THIS tile's 7 glds: the next tile's 7 may stay in flight
    if (kt + 1 < n_ktiles) {
      asm volatile("s_waitcnt vmcnt(7)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    const char* al = a_buf[buf];
    const char* bl = b_buf[buf];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int kk = ks * 32 + (lane >> 4) * 8;
      bf16x8 a_frag[8], b_frag[3];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int row = wm * 128 + i * 16 + (lane & 15);
        const int byte = (kk * 2) ^ ((row & 7) << 4);
        a_frag[i] = *reinterpret_cast<const bf16x8*>(al + row * (kBK * 2) + byte);
      }
#pragma unroll
      for (int j = 0; j < 3; ++j) {
        const int col = wn * 48 + j * 16 + (lane & 15);
        const int byte = (kk * 2) ^ ((col & 7) << 4);
        b_frag[j] = *reinterpret_cast<const bf16x8*>(bl + col * (kBK * 2) + byte);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 3; ++j) acc[i][j] = mfma16g(a_frag[i], b_frag[j], acc[i][j]);
    }
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (kt + 2 < n_ktiles) stage(kt + 2, buf);
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 3; ++j) {
#pragma unroll
      for (int r2 = 0; r2 < 4; ++r2) {
        const int row = row0 + wm * 128 + i * 16 + (lane >> 4) * 4 + r2;
        const int col = n0 + wn * 48 + j * 16 + (lane & 15);
        if (row < row_end && col < N) {
          out[(int64_t)row * N + col] = (bf16_t)acc[i][j][r2];
        }
      }
    }
  }
}

// 8-phase counted-vmcnt gmm_nt (the guide's verified 256-wide template
// adapted to grouped ragged rows), templated on the n-tile width BN8
// (256 or 192 so the production shapes N=576/768/1536/2048 tile exactly).
// 256(M) x BN8(N) tile, BK=64, 8 waves as 2(M) x 4(N), per-wave
// 128 x BN8/4 output. LDS holds exactly TWO k-tiles
// (2 x (A 32KB + B BN8*128B)); staging is global_load_lds_dwordx4:
// A in 128-row halves (2 glds/thread), B in 64-row chunks (1 glds/thread),
// issued inside the compute phases so 2-3 staging units stay in flight
// behind ONE counted s_waitcnt vmcnt(NJ) per k-tile (NJ = BN8/64).
// Each k-tile runs as 4 phases: phase p ds_reads the A fragments of
// C-quadrant p (i in {2p, 2p+1}; B fragments for the whole k-tile are read
// once at phase 0 and live in registers), stages its share of upcoming
// tiles, raw-barriers, and issues 4*NJ MFMAs.
// Death/arrival schedule (slot parity = kt & 1):
//   A(kt+1) goes to the OTHER slot whose tenant A(kt-1) is fully dead
//     -> staged at p0, p1 of kt (one half each)
//   B(kt) region (this slot) dies after phase 0 (all B frags read there)
//     -> B chunks of kt+2 staged at p1..p3
// Per-thread issue order makes the boundary wait vmcnt(NJ): the NJ
// outstanding loads are B chunks of kt+1; everything older has landed.
template <int BN8>
__global__ __launch_bounds__(512, 1) void gmm_nt_8phase_kernel(
    const bf16_t* __restrict__ a,    // (T, K)
    const bf16_t* __restrict__ w,    // (E, N, K)
    bf16_t* __restrict__ out,        // (T, N)
    const int* __restrict__ row_off,
    const int* __restrict__ mtile_pref,
    int E, int K, int N, int n_tiles) {
  constexpr int NJ = BN8 / 64;           // B frags per wave = B chunks
  constexpr int SLOT = (256 + BN8) * 128;  // bytes per k-tile slot
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // slot s (s = kt & 1): A at smem + s*SLOT (32 KB), B right after
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;

  const int nwg = gridDim.x;
  const int xcd = blockIdx.x & 7;
  const int q = nwg >> 3, r = nwg & 7;
  const int wid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                  + (blockIdx.x >> 3);
  const int mt_global = wid / n_tiles;
  int lo = 0, hi = E - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (mtile_pref[mid] <= mt_global) lo = mid; else hi = mid - 1;
  }
  const int e = lo;
  const int m_tile = mt_global - mtile_pref[e];
  const int row0 = row_off[e] + m_tile * kBM;
  const int row_end = row_off[e + 1];
  const int n0 = (wid % n_tiles) * BN8;

  const bf16_t* w_e = w + (int64_t)e * N * K;

  f32x4 acc[8][NJ];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // K % 64 == 32 tail: the last k-tile stages the OVERLAPPING window
  // [K-64, K) (all loads in bounds, no masking -- glds cannot mask) and the
  // MFMA phases run only its upper half (ks = 1), so columns [K-64, K-32)
  // are not double-counted. Host guarantees K % 32 == 0.
  const int n_ktiles = (K + kBK - 1) / kBK;
  const bool k_tail = (K % kBK) != 0;
  const int last_kt = n_ktiles - 1;

  auto ktile_k0 = [&](int kt) {
    return (k_tail && kt == last_kt) ? K - kBK : kt * kBK;
  };

  // A half (128 rows = 16 KB): 2 glds per thread.
  auto stage_a = [&](int kt, int half) {
    const int k0 = ktile_k0(kt);
    char* base = smem + (size_t)(kt & 1) * SLOT + half * (16 * 1024);
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int chunk = wave + i * 8;           // 16 chunks of 1 KB
      const int l = chunk * 1024 + lane * 16;   // linear LDS byte in half
      const int arow = half * 128 + (l >> 7);   // tile row (128-B rows)
      const int colb = (l & 127) ^ ((arow & 7) << 4);  // source pre-swizzle
      const int64_t sr = min(row0 + arow, row_end - 1);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const uint32_t*>(
              reinterpret_cast<const char*>(a) + (sr * (int64_t)K + k0) * 2 + colb),
          reinterpret_cast<uint32_t*>(base + l), 16, 0, 0);
    }
  };
  // B chunk (64 rows = 8 KB): 1 glds per thread.
  auto stage_b = [&](int kt, int chunk) {
    const int k0 = ktile_k0(kt);
    char* base = smem + (size_t)(kt & 1) * SLOT + (32 * 1024)
                 + chunk * (8 * 1024);
    const int l = wave * 1024 + lane * 16;
    const int nrow = chunk * 64 + (l >> 7);
    const int colb = (l & 127) ^ ((nrow & 7) << 4);
    const int64_t sn = min(n0 + nrow, N - 1);
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const uint32_t*>(
            reinterpret_cast<const char*>(w_e) + (sn * (int64_t)K + k0) * 2 + colb),
        reinterpret_cast<uint32_t*>(base + l), 16, 0, 0);
  };

  // Prologue: issue order matches the steady-state boundary count.
#pragma unroll
  for (int c = 0; c < NJ; ++c) stage_b(0, c);
  stage_a(0, 0); stage_a(0, 1);
  if (n_ktiles > 1) {
#pragma unroll
    for (int c = 0; c < NJ; ++c) stage_b(1, c);
  }

  const int n_full = k_tail ? n_ktiles - 1 : n_ktiles;
  for (int kt = 0; kt < n_full; ++kt) {
    const char* a_base = smem + (size_t)(kt & 1) * SLOT;
    const char* b_base = a_base + 32 * 1024;

    // k-tile boundary: this tile's staging landed. In steady state the NJ
    // newer loads are B(kt+1); when B(kt+1) was NOT staged (last tile) the
    // newest outstanding loads are THIS tile's A halves — drain fully
    // (vmcnt(NJ) there let the final tile read in-flight A: rare flake
    // caught by the 12x race screen).
    if (kt + 1 < n_ktiles) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(NJ) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    bf16x8 b_frag[NJ][2];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      // -- ds_read this phase's fragments ---------------------------------
      bf16x8 a_frag[2][2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int kk = ks * 32 + (lane >> 4) * 8;
#pragma unroll
        for (int il = 0; il < 2; ++il) {
          const int row = wm * 128 + (2 * p + il) * 16 + (lane & 15);
          const int byte = (kk * 2) ^ ((row & 7) << 4);
          a_frag[il][ks] = *reinterpret_cast<const bf16x8*>(
              a_base + row * (kBK * 2) + byte);
        }
        if (p == 0) {
#pragma unroll
          for (int j = 0; j < NJ; ++j) {
            const int col = wn * (BN8 / 4) + j * 16 + (lane & 15);
            const int byte = (kk * 2) ^ ((col & 7) << 4);
            b_frag[j][ks] = *reinterpret_cast<const bf16x8*>(
                b_base + col * (kBK * 2) + byte);
          }
        }
      }
      // -- stage upcoming tiles into dead rows ----------------------------
      if (p == 0) {
        if (kt + 1 < n_ktiles) stage_a(kt + 1, 0);
      } else if (p == 1) {
        if (kt + 1 < n_ktiles) stage_a(kt + 1, 1);
        if (kt + 2 < n_ktiles) {
          stage_b(kt + 2, 0);
          if (NJ == 4) stage_b(kt + 2, 1);
        }
      } else if (p == 2) {
        if (kt + 2 < n_ktiles) {
          if (NJ == 4) { stage_b(kt + 2, 2); stage_b(kt + 2, 3); }
          else stage_b(kt + 2, 1);
        }
      } else {
        if (NJ == 3 && kt + 2 < n_ktiles) stage_b(kt + 2, 2);
      }
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int il = 0; il < 2; ++il)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[2 * p + il][j] =
                mfma16g(a_frag[il][ks], b_frag[j][ks], acc[2 * p + il][j]);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

  // Peeled tail k-tile (K % 64 == 32): only its upper half (ks = 1)
  // contributes -- the staged window [K-64, K) overlaps the previous tile.
  if (k_tail) {
    const int kt = last_kt;
    const char* a_base = smem + (size_t)(kt & 1) * SLOT;
    const char* b_base = a_base + 32 * 1024;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    bf16x8 b_frag_t[NJ];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      bf16x8 a_frag_t[2];
      const int kk = 32 + (lane >> 4) * 8;
#pragma unroll
      for (int il = 0; il < 2; ++il) {
        const int row = wm * 128 + (2 * p + il) * 16 + (lane & 15);
        const int byte = (kk * 2) ^ ((row & 7) << 4);
        a_frag_t[il] = *reinterpret_cast<const bf16x8*>(
            a_base + row * (kBK * 2) + byte);
      }
      if (p == 0) {
#pragma unroll
        for (int j = 0; j < NJ; ++j) {
          const int col = wn * (BN8 / 4) + j * 16 + (lane & 15);
          const int byte = (kk * 2) ^ ((col & 7) << 4);
          b_frag_t[j] = *reinterpret_cast<const bf16x8*>(
              b_base + col * (kBK * 2) + byte);
        }
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int il = 0; il < 2; ++il)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
          acc[2 * p + il][j] =
              mfma16g(a_frag_t[il], b_frag_t[j], acc[2 * p + il][j]);
      __builtin_amdgcn_s_setprio(0);
    }
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < NJ; ++j) {
#pragma unroll
      for (int r2 = 0; r2 < 4; ++r2) {
        const int row = row0 + wm * 128 + i * 16 + (lane >> 4) * 4 + r2;
        const int col = n0 + wn * (BN8 / 4) + j * 16 + (lane & 15);
        if (row < row_end && col < N) {
          out[(int64_t)row * N + col] = (bf16_t)acc[i][j][r2];
        }
      }
    }
  }
}

// 8-phase counted-vmcnt gmm for the NN layout (dgrad: out = g @ w[e] with
// w stored (E, K, N), n-contiguous rows). Identical pipeline to
// gmm_nt_8phase_kernel; the ONE difference is the B operand: its MFMA
// fragment needs 8 reduction (k) elements at fixed n — a COLUMN of the
// row-major weight. glds cannot transpose, so B stages ROW-major into a
// [K/4][BN/16][4][16] blocked image (16-byte units are 8 consecutive n at
// one k: still lane-linear, still coalesced) and fragments are read with
// ds_read_b64_tr_b16: each 16-lane group reads one 128-B [4 k][16 n]
// block transposed, two reads per fragment (probe: scratch/tr16_probe.py,
// map in gpurun_out/tr16_map.txt).
template <int BN8>
__global__ __launch_bounds__(512, 1) void gmm_nn_8phase_kernel(
    const bf16_t* __restrict__ a,    // (T, K)
    const bf16_t* __restrict__ w,    // (E, K, N)
    bf16_t* __restrict__ out,        // (T, N)
    const int* __restrict__ row_off,
    const int* __restrict__ mtile_pref,
    int E, int K, int N, int n_tiles) {
  constexpr int NJ = BN8 / 64;
  constexpr int SLOT = (256 + BN8) * 128;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;

  const int nwg = gridDim.x;
  const int xcd = blockIdx.x & 7;
  const int q = nwg >> 3, r = nwg & 7;
  const int wid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                  + (blockIdx.x >> 3);
  const int mt_global = wid / n_tiles;
  int lo = 0, hi = E - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (mtile_pref[mid] <= mt_global) lo = mid; else hi = mid - 1;
  }
  const int e = lo;
  const int m_tile = mt_global - mtile_pref[e];
  const int row0 = row_off[e] + m_tile * kBM;
  const int row_end = row_off[e + 1];
  const int n0 = (wid % n_tiles) * BN8;

  const bf16_t* w_e = w + (int64_t)e * K * N;

  f32x4 acc[8][NJ];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int n_ktiles = (K + kBK - 1) / kBK;
  const bool k_tail = (K % kBK) != 0;   // K % 32 == 0 guaranteed by host
  const int last_kt = n_ktiles - 1;

  auto ktile_k0 = [&](int kt) {
    return (k_tail && kt == last_kt) ? K - kBK : kt * kBK;
  };

  // A half (128 rows x 64 k = 16 KB): 2 glds per thread (same as NT).
  auto stage_a = [&](int kt, int half) {
    const int k0 = ktile_k0(kt);
    char* base = smem + (size_t)(kt & 1) * SLOT + half * (16 * 1024);
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int chunk = wave + i * 8;
      const int l = chunk * 1024 + lane * 16;
      const int arow = half * 128 + (l >> 7);
      const int colb = (l & 127) ^ ((arow & 7) << 4);
      const int64_t sr = min(row0 + arow, row_end - 1);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const uint32_t*>(
              reinterpret_cast<const char*>(a) + (sr * (int64_t)K + k0) * 2 + colb),
          reinterpret_cast<uint32_t*>(base + l), 16, 0, 0);
    }
  };
  // B chunk (8 KB of the blocked [64/4][BN8/16][4][16] k-tile image):
  // 1 glds per thread. Image element e: kb = e / (BN8*4), then
  // nb = (e % (BN8*4)) / 64, krow = (e % 64) / 16, ncol = e % 16.
  auto stage_b = [&](int kt, int chunk) {
    const int k0 = ktile_k0(kt);
    char* base = smem + (size_t)(kt & 1) * SLOT + (32 * 1024);
    const int l = chunk * 8192 + wave * 1024 + lane * 16;  // byte offset
    const int el = l >> 1;                                  // bf16 element
    const int kb = el / (BN8 * 4);
    const int rem = el % (BN8 * 4);
    const int nb = rem >> 6;
    const int krow = (rem & 63) >> 4;
    const int ncol = rem & 15;   // 0 or 8 at 16-B granularity
    const int brow = k0 + kb * 4 + krow;
    const int64_t bcol = min((int64_t)(n0 + nb * 16 + ncol), (int64_t)N - 8);
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const uint32_t*>(
            reinterpret_cast<const char*>(w_e) + ((int64_t)brow * N + bcol) * 2),
        reinterpret_cast<uint32_t*>(base + l), 16, 0, 0);
  };

#pragma unroll
  for (int c = 0; c < NJ; ++c) stage_b(0, c);
  stage_a(0, 0); stage_a(0, 1);
  if (n_ktiles > 1) {
#pragma unroll
    for (int c = 0; c < NJ; ++c) stage_b(1, c);
  }

  const int n_full = k_tail ? n_ktiles - 1 : n_ktiles;
  for (int kt = 0; kt < n_full; ++kt) {
    const char* a_base = smem + (size_t)(kt & 1) * SLOT;
    const char* b_base = a_base + 32 * 1024;

    if (kt + 1 < n_ktiles) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(NJ) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    bf16x8 b_frag[NJ][2];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      bf16x8 a_frag[2][2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int kk = ks * 32 + (lane >> 4) * 8;
#pragma unroll
        for (int il = 0; il < 2; ++il) {
          const int row = wm * 128 + (2 * p + il) * 16 + (lane & 15);
          const int byte = (kk * 2) ^ ((row & 7) << 4);
          a_frag[il][ks] = *reinterpret_cast<const bf16x8*>(
              a_base + row * (kBK * 2) + byte);
        }
        if (p == 0) {
#pragma unroll
          for (int j = 0; j < NJ; ++j) {
            const int nb = (wn * (BN8 / 4) + j * 16) >> 4;
            // fragment k-range kk..kk+8 spans kb blocks 2*(kk/8), +1
            const int kb0 = kk >> 2;  // = (kk/4); kk multiple of 8
            const char* blk0 = b_base + ((size_t)kb0 * (BN8 / 16) + nb) * 128
                               + (lane & 15) * 8;
            const char* blk1 = b_base + ((size_t)(kb0 + 1) * (BN8 / 16) + nb) * 128
                               + (lane & 15) * 8;
            bf16x4 lo4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (__attribute__((address_space(3))) bf16x4*)(blk0));
            bf16x4 hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (__attribute__((address_space(3))) bf16x4*)(blk1));
            bf16x8 f;
#pragma unroll
            for (int t = 0; t < 4; ++t) { f[t] = lo4[t]; f[4 + t] = hi4[t]; }
            b_frag[j][ks] = f;
          }
        }
      }
      if (p == 0) {
        if (kt + 1 < n_ktiles) stage_a(kt + 1, 0);
      } else if (p == 1) {
        if (kt + 1 < n_ktiles) stage_a(kt + 1, 1);
        if (kt + 2 < n_ktiles) {
          stage_b(kt + 2, 0);
          if (NJ == 4) stage_b(kt + 2, 1);
        }
      } else if (p == 2) {
        if (kt + 2 < n_ktiles) {
          if (NJ == 4) { stage_b(kt + 2, 2); stage_b(kt + 2, 3); }
          else stage_b(kt + 2, 1);
        }
      } else {
        if (NJ == 3 && kt + 2 < n_ktiles) stage_b(kt + 2, 2);
      }
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int il = 0; il < 2; ++il)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[2 * p + il][j] =
                mfma16g(a_frag[il][ks], b_frag[j][ks], acc[2 * p + il][j]);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

  if (k_tail) {
    const int kt = last_kt;
    const char* a_base = smem + (size_t)(kt & 1) * SLOT;
    const char* b_base = a_base + 32 * 1024;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    bf16x8 b_frag_t[NJ];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      bf16x8 a_frag_t[2];
      const int kk = 32 + (lane >> 4) * 8;
#pragma unroll
      for (int il = 0; il < 2; ++il) {
        const int row = wm * 128 + (2 * p + il) * 16 + (lane & 15);
        const int byte = (kk * 2) ^ ((row & 7) << 4);
        a_frag_t[il] = *reinterpret_cast<const bf16x8*>(
            a_base + row * (kBK * 2) + byte);
      }
      if (p == 0) {
#pragma unroll
        for (int j = 0; j < NJ; ++j) {
          const int nb = (wn * (BN8 / 4) + j * 16) >> 4;
          const int kb0 = kk >> 2;
          const char* blk0 = b_base + ((size_t)kb0 * (BN8 / 16) + nb) * 128
                             + (lane & 15) * 8;
          const char* blk1 = b_base + ((size_t)(kb0 + 1) * (BN8 / 16) + nb) * 128
                             + (lane & 15) * 8;
          bf16x4 lo4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (__attribute__((address_space(3))) bf16x4*)(blk0));
          bf16x4 hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (__attribute__((address_space(3))) bf16x4*)(blk1));
          bf16x8 f;
#pragma unroll
          for (int t = 0; t < 4; ++t) { f[t] = lo4[t]; f[4 + t] = hi4[t]; }
          b_frag_t[j] = f;
        }
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int il = 0; il < 2; ++il)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
          acc[2 * p + il][j] =
              mfma16g(a_frag_t[il], b_frag_t[j], acc[2 * p + il][j]);
      __builtin_amdgcn_s_setprio(0);
    }
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < NJ; ++j) {
#pragma unroll
      for (int r2 = 0; r2 < 4; ++r2) {
        const int row = row0 + wm * 128 + i * 16 + (lane >> 4) * 4 + r2;
        const int col = n0 + wn * (BN8 / 4) + j * 16 + (lane & 15);
        if (row < row_end && col < N) {
          out[(int64_t)row * N + col] = (bf16_t)acc[i][j][r2];
        }
      }
    }
  }
}

// 8-phase glds + tr16 wgrad: db[e] = a[rows_e]^T @ g[rows_e], out (E, K, N).
// Same block mapping as gmm_db_kernel (expert-order round-robin over
// (e, k-tile, n-tile), ragged row loop in 64-row chunks) but the staging is
// the deep pipeline: BOTH operand chunks stage ROW-major via
// global_load_lds into [rows/4][cols/16][4][16] blocked images and BOTH
// MFMA fragments (reduction = rows) come from ds_read_b64_tr_b16 pairs.
// Rows past the expert's end read from a 16-byte zero buffer (glds cannot
// mask; a per-lane address select costs one v_cndmask, no branch), so pad
// rows contribute exact zeros to the row-sum.
//
// Templated on BOTH output tile dims (256 or 192): the wgrad's output is
// (K x N) and both can quantize badly — the bench's down-projection is
// K = 576 (2.25 tiles of 256 -> 33% wasted MFMA), its gate_up N = 1152
// (4.5 tiles). 192 divides both exactly. Phase count = BK8/64 (k subtiles
// walked two per phase), chunk counts scale with the image sizes, and the
// staging issue order (all a(rt+1) before any g(rt+2)) keeps the counted
// vmcnt waits exact.
template <int BK8, int BN8>
__global__ __launch_bounds__(512, 1) void gmm_db_8phase_kernel(
    const bf16_t* __restrict__ a,   // (T, K)
    const bf16_t* __restrict__ g,   // (T, N)
    bf16_t* __restrict__ db,        // (E, K, N)
    const int* __restrict__ row_off,
    const int* __restrict__ expert_order,
    const bf16_t* __restrict__ zero16,  // >= 8 zero bf16
    int E, int K, int N, int kt_tiles, int nt_tiles) {
  constexpr int KI = BK8 / 32;          // 16-wide k subtiles per wave
  constexpr int NJ = BN8 / 64;          // 16-wide n subtiles per wave
  constexpr int P = KI / 2;             // phases (2 k subtiles each)
  constexpr int ACH = BK8 / 64;         // 8 KB staging chunks per a tile
  constexpr int GCH = BN8 / 64;         // 8 KB staging chunks per g tile
  constexpr int ABYTES = 64 * BK8 * 2;  // blocked-image sizes
  constexpr int GBYTES = 64 * BN8 * 2;
  constexpr int SLOT = ABYTES + GBYTES;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;  // k-half (BK8/2 rows of db)
  const int wn = wave & 3;   // n-quarter (BN8/4 cols)

  const int wid = blockIdx.x;
  const int tiles_per_e = kt_tiles * nt_tiles;
  const int e = expert_order[wid / tiles_per_e];
  const int k0 = ((wid % tiles_per_e) / nt_tiles) * BK8;
  const int n0 = (wid % nt_tiles) * BN8;
  const int r_start = row_off[e];
  const int r_end = row_off[e + 1];

  f32x4 acc[KI][NJ];
#pragma unroll
  for (int i = 0; i < KI; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  if (r_start >= r_end) {
    // empty expert: this block zero-fills its tile so the host can hand us
    // an EMPTY output (a torch::zeros prefill wrote 2*K*N*E bytes per call
    // — ~0.9 GB per wgrad at the bench shape, ~1.5% of the whole step)
    bf16_t* dbe = db + (int64_t)e * K * N;
#pragma unroll
    for (int i = 0; i < KI; ++i)
#pragma unroll
      for (int j = 0; j < NJ; ++j)
#pragma unroll
        for (int r2 = 0; r2 < 4; ++r2) {
          const int krow = k0 + wm * (BK8 / 2) + i * 16 + (lane >> 4) * 4 + r2;
          const int ncol = n0 + wn * (BN8 / 4) + j * 16 + (lane & 15);
          if (krow < K && ncol < N) dbe[(int64_t)krow * N + ncol] = (bf16_t)0;
        }
    return;
  }

  const int n_rtiles = (r_end - r_start + 63) / 64;

  // Stage one 8 KB chunk of an operand tile (width W) for row tile rt:
  // 1 glds per thread. l = byte offset in the blocked image; the inverse
  // mapping recovers (row, col) for the source address.
  auto stage = [&](const bf16_t* src, int64_t stride, int c0, int cmax,
                   int rt, int chunk, int img_off, int W) {
    char* base = smem + (size_t)(rt & 1) * SLOT + img_off;
    const int l = chunk * 8192 + wave * 1024 + lane * 16;  // byte offset
    const int blk = l >> 7;                // 128-B [4][16] block index
    const int inb = l & 127;
    const int rb = blk / (W / 16);
    const int cb = blk % (W / 16);
    const int row = r_start + rt * 64 + rb * 4 + (inb >> 5);
    const int col8 = cb * 16 + (((inb >> 4) & 1) * 8);
    const int64_t col = min((int64_t)(c0 + col8), (int64_t)cmax - 8);
    const char* sp =
        (row < r_end)
            ? reinterpret_cast<const char*>(src) + (row * stride + col) * 2
            : reinterpret_cast<const char*>(zero16);
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const uint32_t*>(sp),
        reinterpret_cast<uint32_t*>(base + l), 16, 0, 0);
  };
  auto stage_a = [&](int rt, int chunk) {
    stage(a, K, k0, K, rt, chunk, 0, BK8);
  };
  auto stage_g = [&](int rt, int chunk) {
    stage(g, N, n0, N, rt, chunk, ABYTES, BN8);
  };

  // tr16 fragment: reduction slice red0..red0+8 at 16 cols [ct*16, ct*16+16)
  auto frag = [&](const char* img, int W, int red0, int ct) -> bf16x8 {
    const int rb0 = red0 >> 2;
    const char* blk0 =
        img + ((size_t)rb0 * (W / 16) + ct) * 128 + (lane & 15) * 8;
    const char* blk1 =
        img + ((size_t)(rb0 + 1) * (W / 16) + ct) * 128 + (lane & 15) * 8;
    bf16x4 lo4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4*)(blk0));
    bf16x4 hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4*)(blk1));
    bf16x8 f;
#pragma unroll
    for (int t = 0; t < 4; ++t) { f[t] = lo4[t]; f[4 + t] = hi4[t]; }
    return f;
  };

  // Prologue mirrors the steady-state issue order (a before g).
#pragma unroll
  for (int c = 0; c < GCH; ++c) stage_g(0, c);
#pragma unroll
  for (int c = 0; c < ACH; ++c) stage_a(0, c);
  if (n_rtiles > 1) {
#pragma unroll
    for (int c = 0; c < GCH; ++c) stage_g(1, c);
  }

  for (int rt = 0; rt < n_rtiles; ++rt) {
    const char* a_img = smem + (size_t)(rt & 1) * SLOT;
    const char* g_img = a_img + ABYTES;

    if (rt + 1 < n_rtiles) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(GCH) : "memory");
    } else {
      // no g(rt+1) staged: the newest outstanding loads are THIS row
      // tile's a chunks — drain fully
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    bf16x8 b_frag[NJ][2];
#pragma unroll
    for (int p = 0; p < P; ++p) {
      bf16x8 a_frag[2][2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int red0 = ks * 32 + (lane >> 4) * 8;
#pragma unroll
        for (int il = 0; il < 2; ++il) {
          const int kct = (wm * (BK8 / 2) + (2 * p + il) * 16) >> 4;
          a_frag[il][ks] = frag(a_img, BK8, red0, kct);
        }
        if (p == 0) {
#pragma unroll
          for (int j = 0; j < NJ; ++j) {
            const int nct = (wn * (BN8 / 4) + j * 16) >> 4;
            b_frag[j][ks] = frag(g_img, BN8, red0, nct);
          }
        }
      }
      // Stage a(rt+1) then g(rt+2), spread evenly over the phases; a fully
      // before g keeps the top-of-iteration vmcnt(GCH) count exact.
      {
        constexpr int TOT = ACH + GCH;
        constexpr int PER = (TOT + P - 1) / P;
#pragma unroll
        for (int s = p * PER; s < (p + 1) * PER && s < TOT; ++s) {
          if (s < ACH) {
            if (rt + 1 < n_rtiles) stage_a(rt + 1, s);
          } else {
            if (rt + 2 < n_rtiles) stage_g(rt + 2, s - ACH);
          }
        }
      }
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int il = 0; il < 2; ++il)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[2 * p + il][j] =
                mfma16g(a_frag[il][ks], b_frag[j][ks], acc[2 * p + il][j]);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

  bf16_t* db_e = db + (int64_t)e * K * N;
#pragma unroll
  for (int i = 0; i < KI; ++i) {
#pragma unroll
    for (int j = 0; j < NJ; ++j) {
#pragma unroll
      for (int r2 = 0; r2 < 4; ++r2) {
        const int krow = k0 + wm * (BK8 / 2) + i * 16 + (lane >> 4) * 4 + r2;
        const int ncol = n0 + wn * (BN8 / 4) + j * 16 + (lane & 15);
        if (krow < K && ncol < N) {
          db_e[(int64_t)krow * N + ncol] = (bf16_t)acc[i][j][r2];
        }
      }
    }
  }
}

// db[e] = a[rows_e]^T @ g[rows_e]: out (E, K, N).
// Grid (ceil(N/256), ceil(K/256), E): 256x256 tiles per workgroup -- 8 waves
// as 2 (k-halves of 128) x 4 (n-quarters of 64) -- looping the expert's rows
// in chunks of 64. Both operands are staged transposed ([dim][row], 128-byte
// LDS rows, XOR swizzle) for k-contiguous fragments. The big tile is what
// matters here: this kernel is HBM-bound on re-reads (a is read once per
// n-tile, g once per k-tile), and 256x256 halves that traffic vs 128x128.
__global__ __launch_bounds__(512, 1) void gmm_db_kernel(
    const bf16_t* __restrict__ a,   // (T, K)
    const bf16_t* __restrict__ g,   // (T, N)
    bf16_t* __restrict__ db,        // (E, K, N)
    const int* __restrict__ row_off,
    const int* __restrict__ expert_order,  // experts sorted by row count desc
    int E, int K, int N, int kt_tiles, int nt_tiles) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* at_lds = reinterpret_cast<bf16_t*>(smem);   // [256 k][64 rows]
  bf16_t* gt_lds = at_lds + 256 * 64;                 // [256 n][64 rows]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;  // k-half (128 rows of db)
  const int wn = wave & 3;   // n-quarter (64 cols)

  // NO XCD grouping here (unlike gmm_nt_kernel): per-tile work scales with
  // the expert's row count, and giving one XCD a contiguous expert-major
  // span concentrates the heavy (sorted-first) experts on it -- measured as
  // an 8-XCD straggler that cost ~15% end-to-end. Round-robin dispatch
  // balances the ragged loop lengths instead. Heavy experts dispatch first
  // so the tail wave holds the small tiles.
  const int wid = blockIdx.x;
  const int tiles_per_e = kt_tiles * nt_tiles;
  const int e = expert_order[wid / tiles_per_e];
  const int k0 = ((wid % tiles_per_e) / nt_tiles) * 256;
  const int n0 = (wid % nt_tiles) * 256;
  const int r_start = row_off[e];
  const int r_end = row_off[e + 1];
  if (r_start >= r_end) return;  // empty expert: db stays zero (host zeros it)

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // T14 staging: each thread owns 4 (4-row x 2-col) blocks per tensor, loaded
  // into registers early (overlapping the previous chunk's MFMAs) with
  // branchless clamped addresses, transposed into LDS with 8-byte writes
  // late. Rows past r_end are zero-filled (they enter the row-sum).
  union u64u {
    uint64_t u;
    ushort s[4];
  };
  u64u a_c0[4], a_c1[4], g_c0[4], g_c1[4];

  auto load_regs = [&](int rt) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = threadIdx.x + it * 512;
      const int d0 = (idx & 127) * 2;
      const int rb = (idx >> 7) * 4;
      const bool k_ok = k0 + d0 + 1 < K;
      const bool n_ok = n0 + d0 + 1 < N;
      const int64_t sk = min((int64_t)(k0 + d0), (int64_t)max(K - 2, 0));
      const int64_t sn = min((int64_t)(n0 + d0), (int64_t)max(N - 2, 0));
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int g_row = rt + rb + i;
        const bool row_ok = g_row < r_end;
        const int64_t sr = min((int64_t)g_row, (int64_t)max(r_end - 1, 0));
        const uint32_t pa = *reinterpret_cast<const uint32_t*>(a + sr * K + sk);
        const uint32_t pg = *reinterpret_cast<const uint32_t*>(g + sr * N + sn);
        const uint32_t va = (row_ok && k_ok) ? pa : 0u;
        const uint32_t vg = (row_ok && n_ok) ? pg : 0u;
        a_c0[it].s[i] = (ushort)(va & 0xffffu);
        a_c1[it].s[i] = (ushort)(va >> 16);
        g_c0[it].s[i] = (ushort)(vg & 0xffffu);
        g_c1[it].s[i] = (ushort)(vg >> 16);
      }
    }
  };

  auto store_lds = [&]() {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = threadIdx.x + it * 512;
      const int d0 = (idx & 127) * 2;
      const int rb = (idx >> 7) * 4;
      const int byte0 = (rb * 2) ^ ((d0 & 7) << 4);
      const int byte1 = (rb * 2) ^ (((d0 + 1) & 7) << 4);
      *reinterpret_cast<uint64_t*>(
          reinterpret_cast<char*>(at_lds) + d0 * 128 + byte0) = a_c0[it].u;
      *reinterpret_cast<uint64_t*>(
          reinterpret_cast<char*>(at_lds) + (d0 + 1) * 128 + byte1) = a_c1[it].u;
      *reinterpret_cast<uint64_t*>(
          reinterpret_cast<char*>(gt_lds) + d0 * 128 + byte0) = g_c0[it].u;
      *reinterpret_cast<uint64_t*>(
          reinterpret_cast<char*>(gt_lds) + (d0 + 1) * 128 + byte1) = g_c1[it].u;
    }
  };

  load_regs(r_start);
  store_lds();
  __syncthreads();

  for (int rt = r_start; rt < r_end; rt += 64) {
    if (rt + 64 < r_end) load_regs(rt + 64);  // issue early

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // 64 rows -> 2 k-steps
      const int rr = ks * 32 + (lane >> 4) * 8;
      bf16x8 a_frag[8], g_frag[4];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int krow = wm * 128 + i * 16 + (lane & 15);
        const int byte = (rr * 2) ^ ((krow & 7) << 4);
        a_frag[i] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(at_lds) + krow * (64 * 2) + byte);
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int ncol = wn * 64 + j * 16 + (lane & 15);
        const int byte = (rr * 2) ^ ((ncol & 7) << 4);
        g_frag[j] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(gt_lds) + ncol * (64 * 2) + byte);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = mfma16g(a_frag[i], g_frag[j], acc[i][j]);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    if (rt + 64 < r_end) {
      store_lds();
      __syncthreads();
    }
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kk = k0 + wm * 128 + i * 16 + (lane >> 4) * 4 + r;
        const int nn = n0 + wn * 64 + j * 16 + (lane & 15);
        if (kk < K && nn < N) {
          db[((int64_t)e * K + kk) * N + nn] = (bf16_t)acc[i][j][r];
        }
      }
    }
  }
}

}  // namespace d9d

// ---------------------------------------------------------------------------

// E == 1 (dense KernelLinear) offsets depend only on the row count: cache
// the device tensors so the hot path pays no per-call CPU alloc + H2D
// (the round-1 measured overhead that cancelled the kernel's 600-vs-260
// TF/s win at bench scale). MoE sizes change per routing step: no cache.
static std::map<std::tuple<int64_t, int, int>,
                std::tuple<torch::Tensor, torch::Tensor, int>> g_e1_offsets;
static std::mutex g_e1_offsets_mu;

static std::tuple<torch::Tensor, torch::Tensor, int> build_offsets(
    torch::Tensor batch_sizes, torch::Device device, int tile_m) {
  const int E = batch_sizes.numel();
  if (E == 1) {
    const int64_t rows = batch_sizes.to(torch::kInt64).item<int64_t>();
    const auto key = std::make_tuple(rows, tile_m, (int)device.index());
    std::lock_guard<std::mutex> lock(g_e1_offsets_mu);
    auto it = g_e1_offsets.find(key);
    if (it != g_e1_offsets.end()) return it->second;
    auto ro = torch::tensor({(int)0, (int)rows},
                            torch::dtype(torch::kInt32)).to(device);
    const int tiles = (int)((rows + tile_m - 1) / tile_m);
    auto mp = torch::tensor({0, tiles}, torch::dtype(torch::kInt32)).to(device);
    auto val = std::make_tuple(ro, mp, tiles);
    g_e1_offsets.emplace(key, val);
    return val;
  }
  auto row_off = torch::empty({E + 1}, torch::dtype(torch::kInt32));
  auto mtile_pref = torch::empty({E + 1}, torch::dtype(torch::kInt32));
  auto bs = batch_sizes.to(torch::kInt64);
  const int64_t* p = bs.data_ptr<int64_t>();
  int32_t* ro = row_off.data_ptr<int32_t>();
  int32_t* mp = mtile_pref.data_ptr<int32_t>();
  int rows = 0, tiles = 0;
  for (int e = 0; e < E; ++e) {
    ro[e] = rows;
    mp[e] = tiles;
    rows += (int)p[e];
    tiles += (int)((p[e] + tile_m - 1) / tile_m);
  }
  ro[E] = rows;
  mp[E] = tiles;
  // total tile count returned from the CPU copy: reading it from the device
  // tensor (`mtile_pref[E].item()`) is a D2H sync on EVERY call, which
  // serializes the launch pipeline (measured ~0.5 ms/call end-to-end).
  return {row_off.to(device, /*non_blocking=*/true),
          mtile_pref.to(device, /*non_blocking=*/true), tiles};
}

torch::Tensor gmm(torch::Tensor a, torch::Tensor b, torch::Tensor batch_sizes) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(b.is_cuda() && b.scalar_type() == torch::kBFloat16 && b.is_contiguous());
  TORCH_CHECK(batch_sizes.device().is_cpu());
  const int T = a.size(0), K = a.size(1);
  const int E = b.size(0), N = b.size(2);
  TORCH_CHECK(b.size(1) == K, "gmm K mismatch");

  auto out = torch::empty({(int64_t)T, (int64_t)N}, a.options());
  if (T == 0) return out;
  auto [row_off, mtile_pref, total_mtiles] =
      build_offsets(batch_sizes, a.device(), d9d::kBM);
  if (total_mtiles == 0) return out;
  auto stream = at::hip::getCurrentHIPStream();

  // 8-phase + tr16-fragment path (B stays row-major, staged by glds into
  // the [K/4][BN/16][4][16] blocked image). D9D_GMM_NN_8PHASE=0 opts out.
  static const bool use_nn8 = []() {
    const char* v = getenv("D9D_GMM_NN_8PHASE");
    return v == nullptr || v[0] != '0';
  }();
  if (use_nn8 && K % 32 == 0 && K >= 64 && N % 16 == 0) {
    const int pad256 = ((N + 255) / 256) * 256 - N;
    const int pad192 = ((N + 191) / 192) * 192 - N;
    const bool use256 = pad256 <= pad192;
    const int bn = use256 ? 256 : 192;
    const int n_tiles8 = (N + bn - 1) / bn;
    const dim3 grid8(n_tiles8 * total_mtiles);
    const size_t smem8 = (size_t)2 * (256 + bn) * 128;
    static bool attr_set_nn = false;
    if (!attr_set_nn) {
      hipFuncSetAttribute(
          reinterpret_cast<const void*>(&d9d::gmm_nn_8phase_kernel<256>),
          hipFuncAttributeMaxDynamicSharedMemorySize, 128 * 1024);
      hipFuncSetAttribute(
          reinterpret_cast<const void*>(&d9d::gmm_nn_8phase_kernel<192>),
          hipFuncAttributeMaxDynamicSharedMemorySize, 112 * 1024);
      attr_set_nn = true;
    }
    if (use256) {
      hipLaunchKernelGGL(d9d::gmm_nn_8phase_kernel<256>, grid8, dim3(512),
                         smem8, stream,
                         reinterpret_cast<const __bf16*>(a.data_ptr()),
                         reinterpret_cast<const __bf16*>(b.data_ptr()),
                         reinterpret_cast<__bf16*>(out.data_ptr()),
                         row_off.data_ptr<int>(), mtile_pref.data_ptr<int>(),
                         E, K, N, n_tiles8);
    } else {
      hipLaunchKernelGGL(d9d::gmm_nn_8phase_kernel<192>, grid8, dim3(512),
                         smem8, stream,
                         reinterpret_cast<const __bf16*>(a.data_ptr()),
                         reinterpret_cast<const __bf16*>(b.data_ptr()),
                         reinterpret_cast<__bf16*>(out.data_ptr()),
                         row_off.data_ptr<int>(), mtile_pref.data_ptr<int>(),
                         E, K, N, n_tiles8);
    }
    return out;
  }

  const int n_tiles = (N + d9d::kBN - 1) / d9d::kBN;
  const dim3 grid(n_tiles * total_mtiles);
  const size_t smem = (d9d::kBM * d9d::kBK + d9d::kBN * d9d::kBK) * sizeof(__bf16);
  hipLaunchKernelGGL(d9d::gmm_kernel, grid, dim3(512), smem, stream,
                     reinterpret_cast<const __bf16*>(a.data_ptr()),
                     reinterpret_cast<const __bf16*>(b.data_ptr()),
                     reinterpret_cast<__bf16*>(out.data_ptr()),
                     row_off.data_ptr<int>(), mtile_pref.data_ptr<int>(),
                     E, K, N, n_tiles);
  return out;
}


torch::Tensor gmm_nt(torch::Tensor a, torch::Tensor w, torch::Tensor batch_sizes) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 && w.is_contiguous());
  TORCH_CHECK(batch_sizes.device().is_cpu());
  const int T = a.size(0), K = a.size(1);
  const int E = w.size(0), N = w.size(1);
  TORCH_CHECK(w.size(2) == K, "gmm_nt K mismatch");

  auto out = torch::empty({(int64_t)T, (int64_t)N}, a.options());
  if (T == 0) return out;
  auto [row_off, mtile_pref, total_mtiles] =
      build_offsets(batch_sizes, a.device(), d9d::kBM);
  if (total_mtiles == 0) return out;

  auto stream = at::hip::getCurrentHIPStream();
  // 8-phase counted-vmcnt schedule (glds half-tile staging interleaved with
  // the MFMA quadrants, one vmcnt(4) per k-tile): the phase-level interleave
  // the plain 2-buffer glds kernel below was missing. Opt out with
  // D9D_GMM_8PHASE=0.
  static const bool use_8phase = []() {
    const char* v = getenv("D9D_GMM_8PHASE");
    return v == nullptr || v[0] != '0';
  }();
  if (use_8phase && K % 32 == 0 && K >= 64) {
    // pick the n-tile with the least padded waste (ties -> measured faster).
    // Production shapes: 576 -> 192 exact (654 vs 552 TF/s), 768 -> both
    // exact (256 faster), 1536/2048 -> 256. D9D_GMM_NT_BN forces a tile.
    static const int force_bn = []() {
      const char* v = getenv("D9D_GMM_NT_BN");
      return v ? atoi(v) : 0;
    }();
    const int pad256 = ((N + 255) / 256) * 256 - N;
    const int pad192 = ((N + 191) / 192) * 192 - N;
    bool use256 = pad256 <= pad192;  // ties -> 256 (measured: 729 vs 655
                                     // TF/s at N=768, 807 vs 742 at 1536)
    if (force_bn == 256) use256 = true;
    if (force_bn == 192) use256 = false;
    const int bn = use256 ? 256 : 192;
    const int n_tiles8 = (N + bn - 1) / bn;
    const dim3 grid8(n_tiles8 * total_mtiles);
    const size_t smem8 = (size_t)2 * (256 + bn) * 128;
    static bool attr_set = false;
    if (!attr_set) {
      hipFuncSetAttribute(
          reinterpret_cast<const void*>(&d9d::gmm_nt_8phase_kernel<256>),
          hipFuncAttributeMaxDynamicSharedMemorySize, 128 * 1024);
      hipFuncSetAttribute(
          reinterpret_cast<const void*>(&d9d::gmm_nt_8phase_kernel<192>),
          hipFuncAttributeMaxDynamicSharedMemorySize, 112 * 1024);
      attr_set = true;
    }
    if (use256) {
      hipLaunchKernelGGL(d9d::gmm_nt_8phase_kernel<256>, grid8, dim3(512),
                         smem8, stream,
                         reinterpret_cast<const __bf16*>(a.data_ptr()),
                         reinterpret_cast<const __bf16*>(w.data_ptr()),
                         reinterpret_cast<__bf16*>(out.data_ptr()),
                         row_off.data_ptr<int>(), mtile_pref.data_ptr<int>(),
                         E, K, N, n_tiles8);
    } else {
      hipLaunchKernelGGL(d9d::gmm_nt_8phase_kernel<192>, grid8, dim3(512),
                         smem8, stream,
                         reinterpret_cast<const __bf16*>(a.data_ptr()),
                         reinterpret_cast<const __bf16*>(w.data_ptr()),
                         reinterpret_cast<__bf16*>(out.data_ptr()),
                         row_off.data_ptr<int>(), mtile_pref.data_ptr<int>(),
                         E, K, N, n_tiles8);
    }
    return out;
  }
  const int n_tiles = (N + d9d::kBN - 1) / d9d::kBN;
  const dim3 grid(n_tiles * total_mtiles);
  // plain glds staging measured 526 vs 609 TF/s for the register-T14
  // pipeline: at 2-deep tile prefetch one iteration of MFMAs (~0.3 us)
  // cannot cover the HBM latency, so the counted vmcnt stalls every tile.
  // Kept for experiments via D9D_GMM_GLDS=1.
  static const bool use_glds = []() {
    const char* v = getenv("D9D_GMM_GLDS");
    return v != nullptr && v[0] == '1';
  }();
  if (use_glds && K % 64 == 0) {
    const size_t smem =
        2 * (d9d::kBM * d9d::kBK + d9d::kBN * d9d::kBK) * sizeof(__bf16);
    hipLaunchKernelGGL(d9d::gmm_nt_glds_kernel, grid, dim3(512), smem, stream,
                       reinterpret_cast<const __bf16*>(a.data_ptr()),
                       reinterpret_cast<const __bf16*>(w.data_ptr()),
                       reinterpret_cast<__bf16*>(out.data_ptr()),
                       row_off.data_ptr<int>(), mtile_pref.data_ptr<int>(),
                       E, K, N, n_tiles);
    return out;
  }
  const size_t smem = (d9d::kBM * d9d::kBK + d9d::kBN * d9d::kBK) * sizeof(__bf16);
  hipLaunchKernelGGL(d9d::gmm_nt_kernel, grid, dim3(512), smem, stream,
                     reinterpret_cast<const __bf16*>(a.data_ptr()),
                     reinterpret_cast<const __bf16*>(w.data_ptr()),
                     reinterpret_cast<__bf16*>(out.data_ptr()),
                     row_off.data_ptr<int>(), mtile_pref.data_ptr<int>(),
                     E, K, N, n_tiles);
  return out;
}

torch::Tensor gmm_db(torch::Tensor a, torch::Tensor g, torch::Tensor batch_sizes,
                     int64_t num_experts) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(g.is_cuda() && g.scalar_type() == torch::kBFloat16 && g.is_contiguous());
  const int K = a.size(1), N = g.size(1);
  const int E = (int)num_experts;

  // empty (not zeros): the 8-phase kernel covers every tile, including
  // zero-filling empty experts' tiles; only the legacy fallback needs the
  // prefill (it early-returns on empty experts)
  auto db = torch::empty({(int64_t)E, (int64_t)K, (int64_t)N}, a.options());
  if (a.size(0) == 0) return db;
  auto [row_off, mtile_pref, total_mtiles_unused] =
      build_offsets(batch_sizes, a.device(), d9d::kBM);
  (void)total_mtiles_unused;
  auto order_cpu = torch::argsort(batch_sizes.to(torch::kInt64), /*dim=*/0,
                                  /*descending=*/true).to(torch::kInt32);
  auto expert_order = order_cpu.to(a.device(), /*non_blocking=*/true);

  // wgrad tile choice: prefer the width that divides the dim exactly
  // (the bench's down-projection K = 576 loses 33% of its MFMA to a
  // 256-tile ceiling; 192 divides it exactly — likewise N = 1152)
  const int BK = (K % 256 != 0 && K % 192 == 0) ? 192 : 256;
  const int BN = (N % 256 != 0 && N % 192 == 0) ? 192 : 256;
  const int nt_tiles = (N + BN - 1) / BN, kt_tiles = (K + BK - 1) / BK;
  const dim3 grid(nt_tiles * kt_tiles * E);
  auto stream = at::hip::getCurrentHIPStream();

  // 8-phase glds + tr16 pipeline (both operands row-major staged).
  // D9D_GMM_DB_8PHASE=0 opts back into the scatter-staged kernel.
  static const bool use_db8 = []() {
    const char* v = getenv("D9D_GMM_DB_8PHASE");
    return v == nullptr || v[0] != '0';
  }();
  // K/N must be multiples of 8: a 16-byte staging unit must never straddle
  // the tensor edge (the clamp would shift VALID columns' data — caught by
  // the ragged K=100 parity case).
  if (use_db8 && K >= 8 && N >= 8 && K % 8 == 0 && N % 8 == 0) {
    // cached 16-byte zero source for ragged row tails (glds cannot mask)
    static std::map<int, torch::Tensor> zeros_by_dev;
    torch::Tensor z;
    {
      std::lock_guard<std::mutex> lock(g_e1_offsets_mu);
      auto it = zeros_by_dev.find((int)a.device().index());
      if (it == zeros_by_dev.end()) {
        z = torch::zeros({8}, a.options());
        zeros_by_dev.emplace((int)a.device().index(), z);
      } else {
        z = it->second;
      }
    }
    static bool attr_set_db = false;
    if (!attr_set_db) {
#define DB_ATTR(BK8, BN8)                                                     \
      hipFuncSetAttribute(                                                    \
          reinterpret_cast<const void*>(                                      \
              &d9d::gmm_db_8phase_kernel<BK8, BN8>),                          \
          hipFuncAttributeMaxDynamicSharedMemorySize,                         \
          2 * 64 * (BK8 + BN8) * 2)
      DB_ATTR(256, 256); DB_ATTR(256, 192); DB_ATTR(192, 256); DB_ATTR(192, 192);
#undef DB_ATTR
      attr_set_db = true;
    }
#define DB_LAUNCH(BK8, BN8)                                                   \
    hipLaunchKernelGGL((d9d::gmm_db_8phase_kernel<BK8, BN8>), grid,           \
                       dim3(512), 2 * 64 * (BK8 + BN8) * 2, stream,           \
                       reinterpret_cast<const __bf16*>(a.data_ptr()),         \
                       reinterpret_cast<const __bf16*>(g.data_ptr()),         \
                       reinterpret_cast<__bf16*>(db.data_ptr()),              \
                       row_off.data_ptr<int>(), expert_order.data_ptr<int>(), \
                       reinterpret_cast<const __bf16*>(z.data_ptr()),         \
                       E, K, N, kt_tiles, nt_tiles)
    if (BK == 256 && BN == 256) DB_LAUNCH(256, 256);
    else if (BK == 256) DB_LAUNCH(256, 192);
    else if (BN == 256) DB_LAUNCH(192, 256);
    else DB_LAUNCH(192, 192);
#undef DB_LAUNCH
    return db;
  }

  db.zero_();  // legacy kernel early-returns on empty experts
  const int nt256 = (N + 255) / 256, kt256 = (K + 255) / 256;
  const dim3 grid256(nt256 * kt256 * E);
  const size_t smem = (2 * 256 * 64) * sizeof(__bf16);
  hipLaunchKernelGGL(d9d::gmm_db_kernel, grid256, dim3(512), smem, stream,
                     reinterpret_cast<const __bf16*>(a.data_ptr()),
                     reinterpret_cast<const __bf16*>(g.data_ptr()),
                     reinterpret_cast<__bf16*>(db.data_ptr()),
                     row_off.data_ptr<int>(), expert_order.data_ptr<int>(),
                     E, K, N, kt256, nt256);
  return db;
}
