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
