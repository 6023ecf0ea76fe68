// RMSNorm forward/backward for CDNA4 (gfx950).
//
// MI355X-native design (replaces the reference's Triton kernels,
// d9d/kernel/normalization/rms/op.py): memory-bound row kernels, bf16 I/O
// vectorized 16 B/lane, fp32 math, whole row staged in LDS so HBM traffic is
// one read + one write. Backward uses per-block fp32 dw accumulation in LDS
// followed by global fp32 atomics (fast on CDNA4).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace d9d {

// One row per block. BLOCK threads stage the raw bf16 row in LDS, compute
// sum(x^2) in fp32, then normalize out of LDS.
template <int BLOCK>
__global__ void rms_norm_fwd_kernel(
    const ushort* __restrict__ x,  // (M, N) bf16 bits
    const ushort* __restrict__ w,  // (N,) bf16 bits
    ushort* __restrict__ y,        // (M, N)
    float* __restrict__ inv_rms,   // (M,)
    int64_t M, int64_t N, float eps, float w_offset) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  ushort* row_lds = reinterpret_cast<ushort*>(smem_raw);          // N bf16
  float* red_lds = reinterpret_cast<float*>(row_lds + ((N + 7) & ~7ll));

  const int64_t n_vec = N / 8;
  for (int64_t row = blockIdx.x; row < M; row += gridDim.x) {
    const ushort* xrow = x + row * N;
    float sumsq = 0.f;
    for (int64_t v = threadIdx.x; v < n_vec; v += BLOCK) {
      Bf16x8 pack;
      pack.u = *reinterpret_cast<const ushort8v*>(xrow + v * 8);
      *reinterpret_cast<ushort8v*>(row_lds + v * 8) = pack.u;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_bits_to_f32(pack.s[j]);
        sumsq += f * f;
      }
    }
    // scalar tail
    for (int64_t i = n_vec * 8 + threadIdx.x; i < N; i += BLOCK) {
      ushort bits = xrow[i];
      row_lds[i] = bits;
      float f = bf16_bits_to_f32(bits);
      sumsq += f * f;
    }
    __syncthreads();
    sumsq = block_reduce_sum<BLOCK>(sumsq, red_lds);
    const float inv = rsqrtf(sumsq / static_cast<float>(N) + eps);
    if (threadIdx.x == 0) inv_rms[row] = inv;

    ushort* yrow = y + row * N;
    for (int64_t v = threadIdx.x; v < n_vec; v += BLOCK) {
      Bf16x8 xin, win, out;
      xin.u = *reinterpret_cast<const ushort8v*>(row_lds + v * 8);
      win.u = *reinterpret_cast<const ushort8v*>(w + v * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = bf16_bits_to_f32(xin.s[j]);
        float wf = bf16_bits_to_f32(win.s[j]) + w_offset;
        out.s[j] = f32_to_bf16_rne(xf * inv * wf);
      }
      *reinterpret_cast<ushort8v*>(yrow + v * 8) = out.u;
    }
    for (int64_t i = n_vec * 8 + threadIdx.x; i < N; i += BLOCK) {
      float xf = bf16_bits_to_f32(row_lds[i]);
      float wf = bf16_bits_to_f32(w[i]) + w_offset;
      yrow[i] = f32_to_bf16_rne(xf * inv * wf);
    }
    __syncthreads();
  }
}

// Small-N specialization (e.g. per-head QK norm, N <= 512): one WAVE per row,
// row in registers, no LDS round-trip. Per-lane elements are CONTIGUOUS
// (lane*VPL + j) so loads/stores vectorize to 4-16 B per lane.
template <int BLOCK, int VPL>  // VPL = values per lane (N <= 64*VPL)
__global__ void rms_norm_fwd_smalln_kernel(
    const ushort* __restrict__ x,
    const ushort* __restrict__ w,
    ushort* __restrict__ y,
    float* __restrict__ inv_rms,
    int64_t M, int64_t N, float eps, float w_offset) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t waves_total = (int64_t)gridDim.x * (BLOCK / 64);
  const int64_t base = (int64_t)lane * VPL;
  const bool full = (VPL >= 2) && (base + VPL <= N);

  float wvals[VPL];
#pragma unroll
  for (int j = 0; j < VPL; ++j) {
    wvals[j] = (base + j < N) ? bf16_bits_to_f32(w[base + j]) + w_offset : 0.f;
  }
  for (int64_t row = blockIdx.x * (BLOCK / 64) + wave; row < M; row += waves_total) {
    const ushort* xrow = x + row * N;
    ushort bits[VPL];
    if (full) {
#pragma unroll
      for (int j = 0; j + 1 < VPL; j += 2) {
        const uint32_t p = *reinterpret_cast<const uint32_t*>(xrow + base + j);
        bits[j] = (ushort)(p & 0xffffu);
        bits[j + 1] = (ushort)(p >> 16);
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPL; ++j) {
        bits[j] = (base + j < N) ? xrow[base + j] : (ushort)0;
      }
    }
    float vals[VPL];
    float sumsq = 0.f;
#pragma unroll
    for (int j = 0; j < VPL; ++j) {
      vals[j] = bf16_bits_to_f32(bits[j]);
      sumsq += vals[j] * vals[j];
    }
    sumsq = wave_reduce_sum(sumsq);
    const float inv = rsqrtf(sumsq / (float)N + eps);
    if (lane == 0) inv_rms[row] = inv;
    ushort* yrow = y + row * N;
    if (full) {
#pragma unroll
      for (int j = 0; j + 1 < VPL; j += 2) {
        const uint32_t p =
            (uint32_t)f32_to_bf16_rne(vals[j] * inv * wvals[j]) |
            ((uint32_t)f32_to_bf16_rne(vals[j + 1] * inv * wvals[j + 1]) << 16);
        *reinterpret_cast<uint32_t*>(yrow + base + j) = p;
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPL; ++j) {
        if (base + j < N) yrow[base + j] = f32_to_bf16_rne(vals[j] * inv * wvals[j]);
      }
    }
  }
}

template <int BLOCK, int VPL>
__global__ void rms_norm_bwd_smalln_kernel(
    const ushort* __restrict__ x,
    const ushort* __restrict__ w,
    const ushort* __restrict__ dy,
    const float* __restrict__ inv_rms,
    ushort* __restrict__ dx,
    float* __restrict__ dw,
    int64_t M, int64_t N, float w_offset) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t waves_total = (int64_t)gridDim.x * (BLOCK / 64);
  const int64_t base = (int64_t)lane * VPL;
  const bool full = (VPL >= 2) && (base + VPL <= N);

  float wvals[VPL], dw_acc[VPL];
#pragma unroll
  for (int j = 0; j < VPL; ++j) {
    wvals[j] = (base + j < N) ? bf16_bits_to_f32(w[base + j]) + w_offset : 0.f;
    dw_acc[j] = 0.f;
  }
  const float inv_n = 1.f / (float)N;
  for (int64_t row = blockIdx.x * (BLOCK / 64) + wave; row < M; row += waves_total) {
    const ushort* xrow = x + row * N;
    const ushort* grow = dy + row * N;
    const float inv = inv_rms[row];
    ushort xb[VPL], gb[VPL];
    if (full) {
#pragma unroll
      for (int j = 0; j + 1 < VPL; j += 2) {
        const uint32_t px = *reinterpret_cast<const uint32_t*>(xrow + base + j);
        const uint32_t pg = *reinterpret_cast<const uint32_t*>(grow + base + j);
        xb[j] = (ushort)(px & 0xffffu); xb[j + 1] = (ushort)(px >> 16);
        gb[j] = (ushort)(pg & 0xffffu); gb[j + 1] = (ushort)(pg >> 16);
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPL; ++j) {
        xb[j] = (base + j < N) ? xrow[base + j] : (ushort)0;
        gb[j] = (base + j < N) ? grow[base + j] : (ushort)0;
      }
    }
    float xh[VPL], g[VPL];
    float s = 0.f;
#pragma unroll
    for (int j = 0; j < VPL; ++j) {
      xh[j] = bf16_bits_to_f32(xb[j]) * inv;
      g[j] = bf16_bits_to_f32(gb[j]);
      s += g[j] * wvals[j] * xh[j];
    }
    s = wave_reduce_sum(s) * inv_n;
    ushort* dxrow = dx + row * N;
    if (full) {
#pragma unroll
      for (int j = 0; j + 1 < VPL; j += 2) {
        const uint32_t p =
            (uint32_t)f32_to_bf16_rne(inv * (g[j] * wvals[j] - xh[j] * s)) |
            ((uint32_t)f32_to_bf16_rne(inv * (g[j + 1] * wvals[j + 1] - xh[j + 1] * s)) << 16);
        *reinterpret_cast<uint32_t*>(dxrow + base + j) = p;
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPL; ++j) {
        if (base + j < N) {
          dxrow[base + j] = f32_to_bf16_rne(inv * (g[j] * wvals[j] - xh[j] * s));
        }
      }
    }
#pragma unroll
    for (int j = 0; j < VPL; ++j) dw_acc[j] += g[j] * xh[j];
  }
  // one global atomic per (wave, element)
#pragma unroll
  for (int j = 0; j < VPL; ++j) {
    if (base + j < N) atomicAdd(dw + base + j, dw_acc[j]);
  }
}


// Head-norm specialization (N <= 128, N % 8 == 0): 16 lanes x bf16x8 per
// row, FOUR rows per wave -- the one-wave-per-row small-N path is latency
// bound on its reduce chain (measured ~1.3 TB/s on the (B*S*H, 128) q/k
// norms); four independent rows per wave hide it.
template <int BLOCK>
__global__ void rms_norm_fwd_qk_kernel(
    const ushort* __restrict__ x,
    const ushort* __restrict__ w,
    ushort* __restrict__ y,
    float* __restrict__ inv_rms,
    int64_t M, int64_t N, float eps, float w_offset) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int sub = lane >> 4;      // row within the wave's group of 4
  const int l16 = lane & 15;
  const int base = l16 * 8;
  // branchless tail handling: loads use a clamped base (always in-bounds),
  // out-of-range elements are zeroed after the load -- a conditional around
  // the load would serialize every iteration behind a vmcnt(0) drain.
  const int safe_base = min(base, max((int)N - 8, 0));
  const bool in_n = base + 8 <= (int)N;
  const int64_t rows_per_iter = (int64_t)gridDim.x * (BLOCK / 64) * 4;
  const float inv_n = 1.f / (float)N;

  float wv[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    wv[j] = (base + j < N) ? bf16_bits_to_f32(w[base + j]) + w_offset : 0.f;
  }

  for (int64_t row = ((int64_t)blockIdx.x * (BLOCK / 64) + wave) * 4 + sub;
       row < M; row += rows_per_iter) {
    Bf16x8 xb;
    xb.u = *reinterpret_cast<const ushort8v*>(x + row * N + safe_base);
    float xv[8], ss = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      xv[j] = in_n ? bf16_bits_to_f32(xb.s[j]) : 0.f;
      ss += xv[j] * xv[j];
    }
#pragma unroll
    for (int off = 1; off < 16; off <<= 1) ss += __shfl_xor(ss, off, 64);
    const float inv = __frsqrt_rn(ss * inv_n + eps);
    Bf16x8 yb;
#pragma unroll
    for (int j = 0; j < 8; ++j) yb.s[j] = f32_to_bf16_rne(xv[j] * inv * wv[j]);
    if (in_n) *reinterpret_cast<ushort8v*>(y + row * N + base) = yb.u;
    if (l16 == 0) inv_rms[row] = inv;
  }
}

template <int BLOCK>
__global__ void rms_norm_bwd_qk_kernel(
    const ushort* __restrict__ x,
    const ushort* __restrict__ w,
    const ushort* __restrict__ dy,
    const float* __restrict__ inv_rms,
    ushort* __restrict__ dx,
    float* __restrict__ dw_part,  // (gridDim.x, N) fp32 per-block partials
    int64_t M, int64_t N, float w_offset) {
  __shared__ float dw_lds[128];
  for (int i = threadIdx.x; i < N; i += BLOCK) dw_lds[i] = 0.f;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int sub = lane >> 4;
  const int l16 = lane & 15;
  const int base = l16 * 8;
  const int safe_base = min(base, max((int)N - 8, 0));  // see fwd_qk
  const bool in_n = base + 8 <= (int)N;
  const int64_t rows_per_iter = (int64_t)gridDim.x * (BLOCK / 64) * 4;
  const float inv_n = 1.f / (float)N;

  float wv[8], dw_acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    wv[j] = (base + j < N) ? bf16_bits_to_f32(w[base + j]) + w_offset : 0.f;
    dw_acc[j] = 0.f;
  }

  for (int64_t row = ((int64_t)blockIdx.x * (BLOCK / 64) + wave) * 4 + sub;
       row < M; row += rows_per_iter) {
    Bf16x8 xb, gb;
    xb.u = *reinterpret_cast<const ushort8v*>(x + row * N + safe_base);
    gb.u = *reinterpret_cast<const ushort8v*>(dy + row * N + safe_base);
    const float inv = inv_rms[row];
    float xh[8], g[8], s = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      xh[j] = in_n ? bf16_bits_to_f32(xb.s[j]) * inv : 0.f;
      g[j] = in_n ? bf16_bits_to_f32(gb.s[j]) : 0.f;
      s += g[j] * wv[j] * xh[j];
    }
#pragma unroll
    for (int off = 1; off < 16; off <<= 1) s += __shfl_xor(s, off, 64);
    s *= inv_n;
    Bf16x8 db;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      db.s[j] = f32_to_bf16_rne(inv * (g[j] * wv[j] - xh[j] * s));
      dw_acc[j] += g[j] * xh[j];
    }
    if (in_n) *reinterpret_cast<ushort8v*>(dx + row * N + base) = db.u;
  }
  // fold the four row-groups (lanes l16, l16+16, ...), then fold waves in
  // LDS and write ONE fp32 partial row per block -- global atomics on 128
  // shared addresses from every wave serialize into milliseconds.
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    dw_acc[j] += __shfl_xor(dw_acc[j], 16, 64);
    dw_acc[j] += __shfl_xor(dw_acc[j], 32, 64);
  }
  if (lane < 16) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (base + j < N) atomicAdd(&dw_lds[base + j], dw_acc[j]);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < N; i += BLOCK) {
    dw_part[(int64_t)blockIdx.x * N + i] = dw_lds[i];
  }
}

// dw[col] = sum over blocks of dw_part[b][col]: one 256-thread block per
// column, tree-reduced (a thread-per-column loop leaves one wave walking
// the whole partials matrix serially).
__global__ void dw_part_reduce_kernel(
    const float* __restrict__ dw_part, float* __restrict__ dw,
    int rows, int64_t N) {
  __shared__ float red[4];
  const int64_t col = blockIdx.x;
  float acc = 0.f;
  for (int r = threadIdx.x; r < rows; r += blockDim.x) {
    acc += dw_part[(int64_t)r * N + col];
  }
  acc = wave_reduce_sum(acc);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    dw[col] += red[0] + red[1] + red[2] + red[3];
  }
}

// Persistent backward: each block walks rows with stride gridDim, keeps a
// per-block fp32 dw accumulator in LDS, and atomically adds it to the global
// fp32 dw buffer once at the end.
template <int BLOCK>
__global__ void rms_norm_bwd_kernel(
    const ushort* __restrict__ x,   // (M, N)
    const ushort* __restrict__ w,   // (N,)
    const ushort* __restrict__ dy,  // (M, N)
    const float* __restrict__ inv_rms,  // (M,)
    ushort* __restrict__ dx,        // (M, N)
    float* __restrict__ dw,         // (N,) fp32, pre-zeroed
    int64_t M, int64_t N, float w_offset) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* dw_lds = reinterpret_cast<float*>(smem_raw);  // N fp32
  float* red_lds = dw_lds + N;                         // 16 fp32

  for (int64_t i = threadIdx.x; i < N; i += BLOCK) dw_lds[i] = 0.f;
  __syncthreads();

  const int64_t n_vec = N / 8;
  const float inv_n = 1.f / static_cast<float>(N);
  for (int64_t row = blockIdx.x; row < M; row += gridDim.x) {
    const ushort* xrow = x + row * N;
    const ushort* grow = dy + row * N;
    const float inv = inv_rms[row];

    // Pass 1: s = sum(dy * w * x_hat); accumulate dw += dy * x_hat.
    float s = 0.f;
    for (int64_t v = threadIdx.x; v < n_vec; v += BLOCK) {
      Bf16x8 xin, win, gin;
      xin.u = *reinterpret_cast<const ushort8v*>(xrow + v * 8);
      win.u = *reinterpret_cast<const ushort8v*>(w + v * 8);
      gin.u = *reinterpret_cast<const ushort8v*>(grow + v * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = bf16_bits_to_f32(xin.s[j]) * inv;
        float g = bf16_bits_to_f32(gin.s[j]);
        float wf = bf16_bits_to_f32(win.s[j]) + w_offset;
        s += g * wf * xh;
        dw_lds[v * 8 + j] += g * xh;  // block-private; no atomics needed yet
      }
    }
    for (int64_t i = n_vec * 8 + threadIdx.x; i < N; i += BLOCK) {
      float xh = bf16_bits_to_f32(xrow[i]) * inv;
      float g = bf16_bits_to_f32(grow[i]);
      float wf = bf16_bits_to_f32(w[i]) + w_offset;
      s += g * wf * xh;
      dw_lds[i] += g * xh;
    }
    __syncthreads();
    s = block_reduce_sum<BLOCK>(s, red_lds) * inv_n;

    // Pass 2: dx = inv * (dy * w - x_hat * s).
    ushort* dxrow = dx + row * N;
    for (int64_t v = threadIdx.x; v < n_vec; v += BLOCK) {
      Bf16x8 xin, win, gin, out;
      xin.u = *reinterpret_cast<const ushort8v*>(xrow + v * 8);
      win.u = *reinterpret_cast<const ushort8v*>(w + v * 8);
      gin.u = *reinterpret_cast<const ushort8v*>(grow + v * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = bf16_bits_to_f32(xin.s[j]) * inv;
        float g = bf16_bits_to_f32(gin.s[j]);
        float wf = bf16_bits_to_f32(win.s[j]) + w_offset;
        out.s[j] = f32_to_bf16_rne(inv * (g * wf - xh * s));
      }
      *reinterpret_cast<ushort8v*>(dxrow + v * 8) = out.u;
    }
    for (int64_t i = n_vec * 8 + threadIdx.x; i < N; i += BLOCK) {
      float xh = bf16_bits_to_f32(xrow[i]) * inv;
      float g = bf16_bits_to_f32(grow[i]);
      float wf = bf16_bits_to_f32(w[i]) + w_offset;
      dxrow[i] = f32_to_bf16_rne(inv * (g * wf - xh * s));
    }
    __syncthreads();
  }

  // Flush the block-private dw accumulator.
  for (int64_t i = threadIdx.x; i < N; i += BLOCK) {
    atomicAdd(dw + i, dw_lds[i]);
  }
}

// NOTE on dx formula: dx_j = inv * (dy_j * w_j) - x_hat_j * inv * mean_k(dy_k w_k x_hat_k)
//                         = inv * (dy_j w_j - x_hat_j * s)   with s = mean(dy*w*x_hat).

}  // namespace d9d

// ---- host wrappers ----------------------------------------------------------

static void check_bf16_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

std::vector<torch::Tensor> rms_norm_fwd(
    torch::Tensor x, torch::Tensor w, double eps, bool zero_centered) {
  check_bf16_2d(x, "x");
  check_bf16_2d(w, "w");
  const int64_t N = x.size(-1);
  const int64_t M = x.numel() / N;
  TORCH_CHECK(w.numel() == N, "weight size mismatch");

  auto y = torch::empty_like(x);
  auto inv_rms = torch::empty({M}, x.options().dtype(torch::kFloat32));

  constexpr int kBlock = 256;
  auto stream = at::hip::getCurrentHIPStream();
  if (N <= 512) {
    const int grid = static_cast<int>(std::min<int64_t>((M + 3) / 4, 2048));
#define SMALL_FWD(VPL)                                                         \
  hipLaunchKernelGGL((d9d::rms_norm_fwd_smalln_kernel<kBlock, VPL>),           \
                     dim3(grid), dim3(kBlock), 0, stream,                      \
                     reinterpret_cast<const ushort*>(x.data_ptr()),            \
                     reinterpret_cast<const ushort*>(w.data_ptr()),            \
                     reinterpret_cast<ushort*>(y.data_ptr()),                  \
                     inv_rms.data_ptr<float>(), M, N,                          \
                     static_cast<float>(eps), zero_centered ? 1.0f : 0.0f)
    if (N <= 128 && N % 8 == 0) {
      const int qgrid = static_cast<int>(std::min<int64_t>((M + 15) / 16, 4096));
      hipLaunchKernelGGL((d9d::rms_norm_fwd_qk_kernel<kBlock>), dim3(qgrid),
                         dim3(kBlock), 0, stream,
                         reinterpret_cast<const ushort*>(x.data_ptr()),
                         reinterpret_cast<const ushort*>(w.data_ptr()),
                         reinterpret_cast<ushort*>(y.data_ptr()),
                         inv_rms.data_ptr<float>(), M, N, (float)eps,
                         zero_centered ? 1.0f : 0.0f);
      return {y, inv_rms};
    }
    if (N <= 64) SMALL_FWD(1);
    else if (N <= 128) SMALL_FWD(2);
    else if (N <= 256) SMALL_FWD(4);
    else SMALL_FWD(8);
#undef SMALL_FWD
    return {y, inv_rms};
  }
  const int grid = static_cast<int>(std::min<int64_t>(M, 2048));
  const size_t smem = ((N + 7) & ~7ll) * sizeof(ushort) + 16 * sizeof(float);
  TORCH_CHECK(smem <= 160 * 1024, "rms_norm: N too large for LDS staging: ", N);
  hipLaunchKernelGGL(
      (d9d::rms_norm_fwd_kernel<kBlock>), dim3(grid), dim3(kBlock), smem, stream,
      reinterpret_cast<const ushort*>(x.data_ptr()),
      reinterpret_cast<const ushort*>(w.data_ptr()),
      reinterpret_cast<ushort*>(y.data_ptr()),
      inv_rms.data_ptr<float>(),
      M, N, static_cast<float>(eps), zero_centered ? 1.0f : 0.0f);
  return {y, inv_rms};
}

std::vector<torch::Tensor> rms_norm_bwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor dy, torch::Tensor inv_rms,
    bool zero_centered) {
  check_bf16_2d(x, "x");
  check_bf16_2d(w, "w");
  check_bf16_2d(dy, "dy");
  const int64_t N = x.size(-1);
  const int64_t M = x.numel() / N;

  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({N}, x.options().dtype(torch::kFloat32));

  constexpr int kBlock = 256;
  auto stream = at::hip::getCurrentHIPStream();
  if (N <= 512) {
    const int sgrid = static_cast<int>(std::min<int64_t>((M + 3) / 4, 1024));
    const size_t ssmem = 0;
#define SMALL_BWD(VPL)                                                         \
  hipLaunchKernelGGL((d9d::rms_norm_bwd_smalln_kernel<kBlock, VPL>),           \
                     dim3(sgrid), dim3(kBlock), ssmem, stream,                 \
                     reinterpret_cast<const ushort*>(x.data_ptr()),            \
                     reinterpret_cast<const ushort*>(w.data_ptr()),            \
                     reinterpret_cast<const ushort*>(dy.data_ptr()),           \
                     inv_rms.data_ptr<float>(),                                \
                     reinterpret_cast<ushort*>(dx.data_ptr()),                 \
                     dw.data_ptr<float>(), M, N, zero_centered ? 1.0f : 0.0f)
    if (N <= 128 && N % 8 == 0) {
      const int qgrid = static_cast<int>(std::min<int64_t>((M + 15) / 16, 4096));
      auto dw_part = torch::empty({qgrid, N}, x.options().dtype(torch::kFloat32));
      hipLaunchKernelGGL((d9d::rms_norm_bwd_qk_kernel<kBlock>), dim3(qgrid),
                         dim3(kBlock), 0, stream,
                         reinterpret_cast<const ushort*>(x.data_ptr()),
                         reinterpret_cast<const ushort*>(w.data_ptr()),
                         reinterpret_cast<const ushort*>(dy.data_ptr()),
                         inv_rms.data_ptr<float>(),
                         reinterpret_cast<ushort*>(dx.data_ptr()),
                         dw_part.data_ptr<float>(), M, N,
                         zero_centered ? 1.0f : 0.0f);
      hipLaunchKernelGGL(d9d::dw_part_reduce_kernel, dim3((int)N),
                         dim3(256), 0, stream, dw_part.data_ptr<float>(),
                         dw.data_ptr<float>(), qgrid, N);
      return {dx, dw};
    }
    if (N <= 64) SMALL_BWD(1);
    else if (N <= 128) SMALL_BWD(2);
    else if (N <= 256) SMALL_BWD(4);
    else SMALL_BWD(8);
#undef SMALL_BWD
    return {dx, dw};
  }
  const int grid = static_cast<int>(std::min<int64_t>(M, 1024));
  const size_t smem = N * sizeof(float) + 16 * sizeof(float);
  TORCH_CHECK(smem <= 160 * 1024, "rms_norm bwd: N too large for LDS dw: ", N);
  hipLaunchKernelGGL(
      (d9d::rms_norm_bwd_kernel<kBlock>), dim3(grid), dim3(kBlock), smem, stream,
      reinterpret_cast<const ushort*>(x.data_ptr()),
      reinterpret_cast<const ushort*>(w.data_ptr()),
      reinterpret_cast<const ushort*>(dy.data_ptr()),
      inv_rms.data_ptr<float>(),
      reinterpret_cast<ushort*>(dx.data_ptr()),
      dw.data_ptr<float>(),
      M, N, zero_centered ? 1.0f : 0.0f);
  return {dx, dw};
}
