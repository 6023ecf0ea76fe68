// Fused rotary position embedding for CDNA4: rotates q AND k in one launch.
//
// Replaces the eager rotate_half composition (reference applies RoPE with
// ~6 elementwise ops per projection; d9d/module/block/positional/rope.py).
// HALF layout: out_i = x_i*cos_i - x_{i+h}*sin_i; out_{i+h} = x_{i+h}*cos_i
// + x_i*sin_i (cos/sin tables are duplicated over the two halves, so only
// the first half is read). Backward is the inverse rotation (sin -> -sin),
// served by the same kernel.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace d9d {

typedef __bf16 bf16_t;

// One thread rotates TWO adjacent (i, i+half) pairs: 4B loads/stores.
__global__ void rope_rotate_kernel(
    const bf16_t* __restrict__ x,    // (B, S, H, D)
    const float* __restrict__ cos_t, // (B, S, rope_dim) fp32, duplicated halves
    const float* __restrict__ sin_t,
    bf16_t* __restrict__ out,        // (B, S, H, D); pass-through d >= rope_dim
    int64_t total_rows,              // B*S*H
    int H, int D, int rope_dim, float sin_sign) {
  const int half = rope_dim / 2;
  const int64_t pairs_per_row = half / 2;  // thread handles 2 pairs
  const int64_t total = total_rows * pairs_per_row;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += stride) {
    const int64_t row = idx / pairs_per_row;
    const int d0 = (int)(idx % pairs_per_row) * 2;
    const int64_t bs = row / H;  // (b*S + s)
    const bf16_t* xp = x + row * D;
    bf16_t* op = out + row * D;
    const float* cp = cos_t + bs * rope_dim;
    const float* sp = sin_t + bs * rope_dim;

    const uint32_t lo = *reinterpret_cast<const uint32_t*>(xp + d0);
    const uint32_t hi = *reinterpret_cast<const uint32_t*>(xp + d0 + half);
    float x0 = bf16_bits_to_f32((ushort)(lo & 0xffffu));
    float x1 = bf16_bits_to_f32((ushort)(lo >> 16));
    float y0 = bf16_bits_to_f32((ushort)(hi & 0xffffu));
    float y1 = bf16_bits_to_f32((ushort)(hi >> 16));
    const float c0 = cp[d0], c1 = cp[d0 + 1];
    const float s0 = sp[d0] * sin_sign, s1 = sp[d0 + 1] * sin_sign;

    const uint32_t olo =
        (uint32_t)f32_to_bf16_rne(x0 * c0 - y0 * s0) |
        ((uint32_t)f32_to_bf16_rne(x1 * c1 - y1 * s1) << 16);
    const uint32_t ohi =
        (uint32_t)f32_to_bf16_rne(y0 * c0 + x0 * s0) |
        ((uint32_t)f32_to_bf16_rne(y1 * c1 + x1 * s1) << 16);
    *reinterpret_cast<uint32_t*>(op + d0) = olo;
    *reinterpret_cast<uint32_t*>(op + d0 + half) = ohi;
  }
}

}  // namespace d9d

static void rope_launch(torch::Tensor x, torch::Tensor cos_t, torch::Tensor sin_t,
                        torch::Tensor out, double sin_sign) {
  const int D = x.size(-1);
  const int H = x.size(-2);
  const int rope_dim = cos_t.size(-1);
  const int64_t total_rows = x.numel() / D;
  const int64_t work = total_rows * (rope_dim / 4);
  const int grid = (int)std::min<int64_t>((work + 255) / 256, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(d9d::rope_rotate_kernel, dim3(grid), dim3(256), 0, stream,
                     reinterpret_cast<const __bf16*>(x.data_ptr()),
                     cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                     reinterpret_cast<__bf16*>(out.data_ptr()),
                     total_rows, H, D, rope_dim, (float)sin_sign);
}

std::vector<torch::Tensor> rope_qk(
    torch::Tensor q, torch::Tensor k,
    torch::Tensor cos_t, torch::Tensor sin_t, double sin_sign) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(k.scalar_type() == torch::kBFloat16 && k.is_contiguous());
  TORCH_CHECK(cos_t.scalar_type() == torch::kFloat32 && cos_t.is_contiguous());
  const int rope_dim = cos_t.size(-1);
  TORCH_CHECK(rope_dim % 4 == 0, "rope_dim must be a multiple of 4");
  // pass-through dims (rope_dim < D) are preserved by starting from a clone
  auto q_out = rope_dim == q.size(-1) ? torch::empty_like(q) : q.clone();
  auto k_out = rope_dim == k.size(-1) ? torch::empty_like(k) : k.clone();
  rope_launch(q, cos_t, sin_t, q_out, sin_sign);
  rope_launch(k, cos_t, sin_t, k_out, sin_sign);
  return {q_out, k_out};
}
