// MoE token permutation kernels for CDNA4.
//
// Replaces the reference's Triton permute/unpermute stack
// (d9d/kernel/moe/permute_with_probs.py, indices_to_multihot.py) and the ATen
// index_select/index_add fallbacks. Deterministic by construction: the
// combine is a CSR-style gather over each token's K replica rows (flat
// replica r of token t is t*K + j, so the inverse permutation gives the rows
// directly) — no atomics, bitwise-reproducible.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace d9d {

typedef __bf16 bf16_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

// out[r] = src[idx[r]] * (scale ? scale[r] : 1)
__global__ void gather_rows_kernel(
    const bf16_t* __restrict__ src,   // (N, H)
    const int64_t* __restrict__ idx,  // (R,)
    const float* __restrict__ scale,  // (R,) or nullptr
    bf16_t* __restrict__ out,         // (R, H)
    int64_t R, int64_t H) {
  const int64_t r = blockIdx.x;
  if (r >= R) return;
  const int64_t s_row = idx[r];
  const float sc = scale ? scale[r] : 1.0f;
  const bf16_t* sp = src + s_row * H;
  bf16_t* op = out + r * H;
  if (scale) {
    for (int64_t i = threadIdx.x * 8; i + 7 < H; i += blockDim.x * 8) {
      bf16x8 vv = *reinterpret_cast<const bf16x8*>(sp + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) vv[j] = (bf16_t)((float)vv[j] * sc);
      *reinterpret_cast<bf16x8*>(op + i) = vv;
    }
    for (int64_t i = (H / 8) * 8 + threadIdx.x; i < H; i += blockDim.x) {
      op[i] = (bf16_t)((float)sp[i] * sc);
    }
  } else {
    for (int64_t i = threadIdx.x * 8; i + 7 < H; i += blockDim.x * 8) {
      *reinterpret_cast<bf16x8*>(op + i) = *reinterpret_cast<const bf16x8*>(sp + i);
    }
    for (int64_t i = (H / 8) * 8 + threadIdx.x; i < H; i += blockDim.x) {
      op[i] = sp[i];
    }
  }
}

// out[t] = sum_j expert_out[inv[t*K+j]] * (probs ? probs[inv[t*K+j]] : 1)
__global__ void csr_combine_kernel(
    const bf16_t* __restrict__ expert_out,  // (R, H)
    const float* __restrict__ probs,        // (R,) or nullptr
    const int64_t* __restrict__ inv,        // (R,) inverse permutation
    bf16_t* __restrict__ out,               // (T, H)
    int64_t T, int64_t H, int K) {
  const int64_t t = blockIdx.x;
  if (t >= T) return;
  bf16_t* op = out + t * H;
  for (int64_t i = threadIdx.x * 8; i + 7 < H; i += blockDim.x * 8) {
    float acc[8] = {};
    for (int j = 0; j < K; ++j) {
      const int64_t r = inv[t * K + j];
      const float p = probs ? probs[r] : 1.0f;
      const bf16x8 vv = *reinterpret_cast<const bf16x8*>(expert_out + r * H + i);
#pragma unroll
      for (int q = 0; q < 8; ++q) acc[q] += (float)vv[q] * p;
    }
    bf16x8 ov;
#pragma unroll
    for (int q = 0; q < 8; ++q) ov[q] = (bf16_t)acc[q];
    *reinterpret_cast<bf16x8*>(op + i) = ov;
  }
  for (int64_t i = (H / 8) * 8 + threadIdx.x; i < H; i += blockDim.x) {
    float acc = 0.f;
    for (int j = 0; j < K; ++j) {
      const int64_t r = inv[t * K + j];
      const float p = probs ? probs[r] : 1.0f;
      acc += (float)expert_out[r * H + i] * p;
    }
    op[i] = (bf16_t)acc;
  }
}

// dprobs[r] = dot(grad_out[row_to_token[r]], expert_out[r])  (fp32)
__global__ void row_dot_kernel(
    const bf16_t* __restrict__ grad_out,   // (T, H)
    const bf16_t* __restrict__ expert_out, // (R, H)
    const int64_t* __restrict__ row_to_token,
    float* __restrict__ dprobs,            // (R,)
    int64_t R, int64_t H) {
  const int64_t r = blockIdx.x;
  if (r >= R) return;
  const bf16_t* gp = grad_out + row_to_token[r] * H;
  const bf16_t* ep = expert_out + r * H;
  float acc = 0.f;
  for (int64_t i = threadIdx.x * 8; i + 7 < H; i += 64 * 8) {
    const bf16x8 gv = *reinterpret_cast<const bf16x8*>(gp + i);
    const bf16x8 ev = *reinterpret_cast<const bf16x8*>(ep + i);
#pragma unroll
    for (int q = 0; q < 8; ++q) acc += (float)gv[q] * (float)ev[q];
  }
  for (int64_t i = (H / 8) * 8 + threadIdx.x; i < H; i += 64) {
    acc += (float)gp[i] * (float)ep[i];
  }
  acc = wave_reduce_sum(acc);
  if (threadIdx.x == 0) dprobs[r] = acc;
}

}  // namespace d9d

torch::Tensor moe_gather_rows(
    torch::Tensor src, torch::Tensor idx,
    c10::optional<torch::Tensor> scale) {
  TORCH_CHECK(src.is_cuda() && src.scalar_type() == torch::kBFloat16 && src.is_contiguous());
  TORCH_CHECK(idx.scalar_type() == torch::kInt64);
  const int64_t R = idx.numel(), H = src.size(1);
  auto out = torch::empty({R, H}, src.options());
  if (R == 0) return out;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(
      d9d::gather_rows_kernel, dim3((unsigned)R), dim3(64), 0, stream,
      reinterpret_cast<const __bf16*>(src.data_ptr()),
      idx.data_ptr<int64_t>(),
      scale ? scale->data_ptr<float>() : nullptr,
      reinterpret_cast<__bf16*>(out.data_ptr()), R, H);
  return out;
}

torch::Tensor moe_csr_combine(
    torch::Tensor expert_out, c10::optional<torch::Tensor> probs,
    torch::Tensor inv, int64_t T, int64_t K) {
  TORCH_CHECK(expert_out.is_cuda() && expert_out.scalar_type() == torch::kBFloat16);
  const int64_t H = expert_out.size(1);
  auto out = torch::empty({T, H}, expert_out.options());
  if (T == 0) return out;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(
      d9d::csr_combine_kernel, dim3((unsigned)T), dim3(64), 0, stream,
      reinterpret_cast<const __bf16*>(expert_out.contiguous().data_ptr()),
      probs ? probs->data_ptr<float>() : nullptr,
      inv.data_ptr<int64_t>(),
      reinterpret_cast<__bf16*>(out.data_ptr()), T, H, (int)K);
  return out;
}

torch::Tensor moe_row_dot(
    torch::Tensor grad_out, torch::Tensor expert_out, torch::Tensor row_to_token) {
  const int64_t R = expert_out.size(0), H = expert_out.size(1);
  auto dprobs = torch::empty({R}, grad_out.options().dtype(torch::kFloat32));
  if (R == 0) return dprobs;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(
      d9d::row_dot_kernel, dim3((unsigned)R), dim3(64), 0, stream,
      reinterpret_cast<const __bf16*>(grad_out.contiguous().data_ptr()),
      reinterpret_cast<const __bf16*>(expert_out.contiguous().data_ptr()),
      row_to_token.data_ptr<int64_t>(),
      dprobs.data_ptr<float>(), R, H);
  return dprobs;
}
