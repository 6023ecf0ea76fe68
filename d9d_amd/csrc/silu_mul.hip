// Fused SiLU-mul (SwiGLU activation) forward/backward for CDNA4.
//
// Replaces the reference's Triton kernels (d9d/kernel/swiglu/op.py).
// Pure HBM-bound elementwise: bf16 I/O at 16 B/lane, fp32 math, grid-stride.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace d9d {

D9D_DEVICE float silu_f(float x) {
  const float sig = 1.f / (1.f + __builtin_amdgcn_exp2f(-x * 1.44269504f));
  return x * sig;
}

template <int BLOCK>
__global__ void silu_mul_fwd_kernel(
    const ushort* __restrict__ a,  // gate
    const ushort* __restrict__ b,  // up
    ushort* __restrict__ out,
    int64_t n) {
  const int64_t n_vec = n / 8;
  const int64_t tid = blockIdx.x * static_cast<int64_t>(BLOCK) + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * BLOCK;
  for (int64_t v = tid; v < n_vec; v += stride) {
    Bf16x8 av, bv, ov;
    av.u = *reinterpret_cast<const ushort8v*>(a + v * 8);
    bv.u = *reinterpret_cast<const ushort8v*>(b + v * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      ov.s[j] = f32_to_bf16_rne(
          silu_f(bf16_bits_to_f32(av.s[j])) * bf16_bits_to_f32(bv.s[j]));
    }
    *reinterpret_cast<ushort8v*>(out + v * 8) = ov.u;
  }
  for (int64_t i = n_vec * 8 + tid; i < n; i += stride) {
    out[i] = f32_to_bf16_rne(silu_f(bf16_bits_to_f32(a[i])) * bf16_bits_to_f32(b[i]));
  }
}

template <int BLOCK>
__global__ void silu_mul_bwd_kernel(
    const ushort* __restrict__ a,
    const ushort* __restrict__ b,
    const ushort* __restrict__ g,
    ushort* __restrict__ da,
    ushort* __restrict__ db,
    int64_t n) {
  const int64_t n_vec = n / 8;
  const int64_t tid = blockIdx.x * static_cast<int64_t>(BLOCK) + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * BLOCK;
  for (int64_t v = tid; v < n_vec; v += stride) {
    Bf16x8 av, bv, gv, dav, dbv;
    av.u = *reinterpret_cast<const ushort8v*>(a + v * 8);
    bv.u = *reinterpret_cast<const ushort8v*>(b + v * 8);
    gv.u = *reinterpret_cast<const ushort8v*>(g + v * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float af = bf16_bits_to_f32(av.s[j]);
      const float bf = bf16_bits_to_f32(bv.s[j]);
      const float gf = bf16_bits_to_f32(gv.s[j]);
      const float sig = 1.f / (1.f + __builtin_amdgcn_exp2f(-af * 1.44269504f));
      const float silu = af * sig;
      dav.s[j] = f32_to_bf16_rne(gf * bf * sig * (1.f + af * (1.f - sig)));
      dbv.s[j] = f32_to_bf16_rne(gf * silu);
    }
    *reinterpret_cast<ushort8v*>(da + v * 8) = dav.u;
    *reinterpret_cast<ushort8v*>(db + v * 8) = dbv.u;
  }
  for (int64_t i = n_vec * 8 + tid; i < n; i += stride) {
    const float af = bf16_bits_to_f32(a[i]);
    const float bf = bf16_bits_to_f32(b[i]);
    const float gf = bf16_bits_to_f32(g[i]);
    const float sig = 1.f / (1.f + __builtin_amdgcn_exp2f(-af * 1.44269504f));
    da[i] = f32_to_bf16_rne(gf * bf * sig * (1.f + af * (1.f - sig)));
    db[i] = f32_to_bf16_rne(gf * af * sig);
  }
}


// Packed variant: x (T, 2I) holds [gate | up] from the fused gate_up grouped
// GEMM; out (T, I). Avoids materialising two separate projections and the
// dgrad-sum the autograd graph would otherwise insert.
template <int BLOCK>
__global__ void silu_mul_packed_fwd_kernel(
    const ushort* __restrict__ x,
    ushort* __restrict__ out,
    int64_t rows, int inter) {
  const int64_t n_vec = rows * (inter / 8);
  const int64_t tid = blockIdx.x * static_cast<int64_t>(BLOCK) + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * BLOCK;
  const int iv = inter / 8;
  for (int64_t v = tid; v < n_vec; v += stride) {
    // u32 div: a 64-bit software divide per element was ~the same cost
    // as the whole silu math chain
    const uint32_t t = (uint32_t)v / (uint32_t)iv;
    const int i = (int)((uint32_t)v - t * (uint32_t)iv) * 8;
    const ushort* xp = x + t * (int64_t)(2 * inter);
    Bf16x8 av, bv, ov;
    av.u = *reinterpret_cast<const ushort8v*>(xp + i);
    bv.u = *reinterpret_cast<const ushort8v*>(xp + inter + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      ov.s[j] = f32_to_bf16_rne(
          silu_f(bf16_bits_to_f32(av.s[j])) * bf16_bits_to_f32(bv.s[j]));
    }
    *reinterpret_cast<ushort8v*>(out + t * (int64_t)inter + i) = ov.u;
  }
}

template <int BLOCK>
__global__ void silu_mul_packed_bwd_kernel(
    const ushort* __restrict__ x,
    const ushort* __restrict__ g,   // (T, I)
    ushort* __restrict__ dx,        // (T, 2I)
    int64_t rows, int inter) {
  const int64_t n_vec = rows * (inter / 8);
  const int64_t tid = blockIdx.x * static_cast<int64_t>(BLOCK) + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * BLOCK;
  const int iv = inter / 8;
  for (int64_t v = tid; v < n_vec; v += stride) {
    // u32 div: a 64-bit software divide per element was ~the same cost
    // as the whole silu math chain
    const uint32_t t = (uint32_t)v / (uint32_t)iv;
    const int i = (int)((uint32_t)v - t * (uint32_t)iv) * 8;
    const ushort* xp = x + t * (int64_t)(2 * inter);
    ushort* dxp = dx + t * (int64_t)(2 * inter);
    Bf16x8 av, bv, gv, dav, dbv;
    av.u = *reinterpret_cast<const ushort8v*>(xp + i);
    bv.u = *reinterpret_cast<const ushort8v*>(xp + inter + i);
    gv.u = *reinterpret_cast<const ushort8v*>(g + t * (int64_t)inter + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float af = bf16_bits_to_f32(av.s[j]);
      const float bf = bf16_bits_to_f32(bv.s[j]);
      const float gf = bf16_bits_to_f32(gv.s[j]);
      const float sig = 1.f / (1.f + __builtin_amdgcn_exp2f(-af * 1.44269504f));
      dav.s[j] = f32_to_bf16_rne(gf * bf * sig * (1.f + af * (1.f - sig)));
      dbv.s[j] = f32_to_bf16_rne(gf * af * sig);
    }
    *reinterpret_cast<ushort8v*>(dxp + i) = dav.u;
    *reinterpret_cast<ushort8v*>(dxp + inter + i) = dbv.u;
  }
}

}  // namespace d9d

static int silu_grid(int64_t n_vec, int block) {
  const int64_t blocks = d9d::ceil_div(n_vec, block);
  return static_cast<int>(std::min<int64_t>(blocks, 2048));
}

torch::Tensor silu_mul_fwd(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(b.sizes() == a.sizes() && b.scalar_type() == torch::kBFloat16 && b.is_contiguous());
  auto out = torch::empty_like(a);
  const int64_t n = a.numel();
  if (n == 0) return out;
  constexpr int kBlock = 256;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(
      (d9d::silu_mul_fwd_kernel<kBlock>), dim3(silu_grid(n / 8 + 1, kBlock)),
      dim3(kBlock), 0, stream,
      reinterpret_cast<const ushort*>(a.data_ptr()),
      reinterpret_cast<const ushort*>(b.data_ptr()),
      reinterpret_cast<ushort*>(out.data_ptr()), n);
  return out;
}

std::vector<torch::Tensor> silu_mul_bwd(
    torch::Tensor a, torch::Tensor b, torch::Tensor g) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16 && a.is_contiguous());
  auto da = torch::empty_like(a);
  auto db = torch::empty_like(b);
  const int64_t n = a.numel();
  if (n == 0) return {da, db};
  constexpr int kBlock = 256;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(
      (d9d::silu_mul_bwd_kernel<kBlock>), dim3(silu_grid(n / 8 + 1, kBlock)),
      dim3(kBlock), 0, stream,
      reinterpret_cast<const ushort*>(a.data_ptr()),
      reinterpret_cast<const ushort*>(b.data_ptr()),
      reinterpret_cast<const ushort*>(g.data_ptr()),
      reinterpret_cast<ushort*>(da.data_ptr()),
      reinterpret_cast<ushort*>(db.data_ptr()), n);
  return {da, db};
}


torch::Tensor silu_mul_packed_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  const int64_t rows = x.size(0);
  const int inter = (int)x.size(1) / 2;
  TORCH_CHECK(x.size(1) % 2 == 0 && inter % 8 == 0, "packed silu_mul needs I % 8 == 0");
  auto out = torch::empty({rows, (int64_t)inter}, x.options());
  if (out.numel() == 0) return out;
  constexpr int kBlock = 256;
  auto stream = at::hip::getCurrentHIPStream();
  TORCH_CHECK(rows * (int64_t)(inter / 8) < (int64_t)1 << 31,
              "silu_mul_packed: activation too large for u32 indexing");
  hipLaunchKernelGGL(
      (d9d::silu_mul_packed_fwd_kernel<kBlock>),
      dim3(silu_grid(rows * (inter / 8), kBlock)), dim3(kBlock), 0, stream,
      reinterpret_cast<const ushort*>(x.data_ptr()),
      reinterpret_cast<ushort*>(out.data_ptr()), rows, inter);
  return out;
}

torch::Tensor silu_mul_packed_bwd(torch::Tensor x, torch::Tensor g) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(g.is_contiguous() && g.scalar_type() == torch::kBFloat16);
  const int64_t rows = x.size(0);
  const int inter = (int)x.size(1) / 2;
  auto dx = torch::empty_like(x);
  if (x.numel() == 0) return dx;
  constexpr int kBlock = 256;
  auto stream = at::hip::getCurrentHIPStream();
  TORCH_CHECK(rows * (int64_t)(inter / 8) < (int64_t)1 << 31,
              "silu_mul_packed: activation too large for u32 indexing");
  hipLaunchKernelGGL(
      (d9d::silu_mul_packed_bwd_kernel<kBlock>),
      dim3(silu_grid(rows * (inter / 8), kBlock)), dim3(kBlock), 0, stream,
      reinterpret_cast<const ushort*>(x.data_ptr()),
      reinterpret_cast<const ushort*>(g.data_ptr()),
      reinterpret_cast<ushort*>(dx.data_ptr()), rows, inter);
  return dx;
}
