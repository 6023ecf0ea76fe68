// Fused linear + cross-entropy FORWARD for CDNA4 (gfx950).
//
// Replaces the cut-cross-entropy forward (reference: d9d/kernel/cce/cce.py
// cce_lse_forward_kernel): per token, lse over the full vocab and the target
// logit, WITHOUT materializing the (T, V) logits. The (T,V) GEMM runs on
// MFMA inside the kernel with an online logsumexp.
//
// Geometry: one block = 64 token rows (4 waves x 16-row m-tile). The token
// block's embeddings (64 x K bf16) live in LDS for the whole sweep; the
// classifier streams as (64 vocab x 64 k) tiles, T14-staged (loads issued a
// tile early into registers, written to the alternate LDS buffer after the
// MFMAs). All resident blocks sweep the vocab in the same order, so the
// classifier is served from L2/L3 after the first wavefront of blocks.
//
// K (hidden) is a runtime dim; K <= 1024 fits the 160 KiB LDS (the host
// falls back to the chunked rocBLAS path above that). Backward stays the
// chunked-GEMM path (cce.py) — it is GEMM-shaped and rocBLAS-efficient.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace d9d {

typedef __bf16 bf16_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

D9D_DEVICE f32x4 mfma16c(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

constexpr float kLog2eC = 1.44269504088896340736f;

// KT = k-tile width (128 when K % 128 == 0): wider k-tiles double the MFMA
// work between barriers (16 vs 8 per wave); the classifier tile is
// single-buffered with T14 register staging (the registers are the second
// buffer), like the grouped-GEMM kernels.
template <int KT>
__global__ __launch_bounds__(512, 2) void cce_fwd_kernel(
    const bf16_t* __restrict__ e,   // (T, K)
    const bf16_t* __restrict__ c,   // (V, K)
    const int64_t* __restrict__ targets,  // (T,)
    float* __restrict__ lse_out,          // (T,)
    float* __restrict__ tgt_out,          // (T,) target logit or -inf
    int T, int V, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* e_lds = reinterpret_cast<bf16_t*>(smem);       // [64][K] swizzled
  bf16_t* c_lds = e_lds + 64 * K;                        // [128][KT] swizzled
  constexpr int kSwz = (KT == 64) ? 7 : 15;  // XOR stays within the KT*2 row

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;   // 8 waves: (vocab half) x (4 m-tiles)
  const int m_tile = wave & 3;
  const int vhalf = wave >> 2;
  const int t0 = blockIdx.x * 64;

  const int n_vtiles = (V + 127) / 128;  // 128-wide vocab tiles (2 x 64 halves)
  const int n_ktiles = K / KT;
  const int total_iters = n_vtiles * n_ktiles;

  constexpr int kCRegs = KT / 32;  // (128 * KT) / (512 threads * 8)
  bf16x8 c_reg[kCRegs];
  auto load_c = [&](int it_lin) {
    const int vt = it_lin / n_ktiles;
    const int kt = it_lin % n_ktiles;
#pragma unroll
    for (int i = 0; i < kCRegs; ++i) {
      const int idx = (threadIdx.x + i * 512) * 8;
      const int vrow = idx / KT;
      const int col = idx % KT;
      const int g_v = min(vt * 128 + vrow, V - 1);
      c_reg[i] = *reinterpret_cast<const bf16x8*>(
          c + (int64_t)g_v * K + kt * KT + col);
    }
  };
  auto store_c = [&]() {
#pragma unroll
    for (int i = 0; i < kCRegs; ++i) {
      const int idx = (threadIdx.x + i * 512) * 8;
      const int vrow = idx / KT;
      const int col = idx % KT;
      const int byte = (col * 2) ^ ((vrow & kSwz) << 4);
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(c_lds) + vrow * (KT * 2) + byte) = c_reg[i];
    }
  };

  // ---- stage e block (row-major [64][K], ((row&15)<<4) swizzle) and the
  // first classifier tile; ONE barrier covers both ---------------------------
  for (int idx = threadIdx.x * 8; idx < 64 * K; idx += 512 * 8) {
    const int row = idx / K;
    const int col = idx % K;
    const int g_row = min(t0 + row, T - 1);
    const bf16x8 val = *reinterpret_cast<const bf16x8*>(e + (int64_t)g_row * K + col);
    const int byte = (col * 2) ^ ((row & 15) << 4);
    *reinterpret_cast<bf16x8*>(
        reinterpret_cast<char*>(e_lds) + row * (K * 2) + byte) = val;
  }
  load_c(0);
  store_c();
  __syncthreads();

  const int my_row_local = m_tile * 16 + (lane & 15);

  float m_run[4], l_run[4], tgt[4];
  int tg_row[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
    tgt[r] = -1e30f;
    tg_row[r] = (int)targets[min(t0 + m_tile * 16 + (lane >> 4) * 4 + r, T - 1)];
  }

  f32x4 acc[4];
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) acc[nt] = {0.f, 0.f, 0.f, 0.f};

  for (int it = 0; it < total_iters; ++it) {
    const int vt = it / n_ktiles;
    const int kt = it % n_ktiles;
    if (it + 1 < total_iters) load_c(it + 1);  // issue early, store late

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < KT / 32; ++ks) {
      const int kk = kt * KT + ks * 32 + (lane >> 4) * 8;
      const int ebyte = (kk * 2) ^ ((my_row_local & 15) << 4);
      const bf16x8 ea = *reinterpret_cast<const bf16x8*>(
          reinterpret_cast<char*>(e_lds) + my_row_local * (K * 2) + ebyte);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int vrow = vhalf * 64 + nt * 16 + (lane & 15);
        const int cbyte =
            ((ks * 32 + (lane >> 4) * 8) * 2) ^ ((vrow & kSwz) << 4);
        const bf16x8 cb = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(c_lds) + vrow * (KT * 2) + cbyte);
        acc[nt] = mfma16c(ea, cb, acc[nt]);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    if (kt == n_ktiles - 1) {
      // Logits tile complete. PER-LANE online lse over this lane's own 4
      // columns — the 16-lane merge happens ONCE after the sweep, so the
      // hot epilogue has NO cross-lane shuffles (they were ~2/3 of the
      // per-tile VALU work: 96 shfl_xor per vocab tile).
      float vals[4][4];
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int col = vt * 128 + vhalf * 64 + nt * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          vals[nt][r] = (col < V) ? acc[nt][r] : -1e30f;
        }
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          const int col = vt * 128 + vhalf * 64 + nt * 16 + (lane & 15);
          if (col == (int)tg_row[r]) tgt[r] = fmaxf(tgt[r], vals[nt][r]);
        }
        const float mx = fmaxf(fmaxf(vals[0][r], vals[1][r]),
                               fmaxf(vals[2][r], vals[3][r]));
        const float m_new = fmaxf(m_run[r], mx);
        float add = 0.f;
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          add += __builtin_amdgcn_exp2f((vals[nt][r] - m_new) * kLog2eC);
        }
        l_run[r] = l_run[r] *
                       __builtin_amdgcn_exp2f((m_run[r] - m_new) * kLog2eC) +
                   add;
        m_run[r] = m_new;
      }
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) acc[nt] = {0.f, 0.f, 0.f, 0.f};
    }

    __syncthreads();
    if (it + 1 < total_iters) {
      store_c();
      __syncthreads();
    }
  }

  // ---- merge this wave's 16 col-lanes (once per kernel), then the two
  // vocab halves; write lse + target logit ----------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
#pragma unroll
    for (int off = 1; off < 16; off <<= 1) {
      const float mo = __shfl_xor(m_run[r], off, 64);
      const float lo = __shfl_xor(l_run[r], off, 64);
      const float m2 = fmaxf(m_run[r], mo);
      l_run[r] = l_run[r] * __builtin_amdgcn_exp2f((m_run[r] - m2) * kLog2eC) +
                 lo * __builtin_amdgcn_exp2f((mo - m2) * kLog2eC);
      m_run[r] = m2;
      tgt[r] = fmaxf(tgt[r], __shfl_xor(tgt[r], off, 64));
    }
  }
  // scratch (reuses c_lds): per half, per row: lse and tgt.
  float* half_lse = reinterpret_cast<float*>(c_lds);          // [2][64]
  float* half_tgt = half_lse + 2 * 64;                        // [2][64]
  __syncthreads();
  if ((lane & 15) == 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row_l = m_tile * 16 + (lane >> 4) * 4 + r;
      half_lse[vhalf * 64 + row_l] = m_run[r] + __logf(fmaxf(l_run[r], 1e-30f));
      half_tgt[vhalf * 64 + row_l] = tgt[r];
    }
  }
  __syncthreads();
  if (vhalf == 0 && (lane & 15) == 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row_l = m_tile * 16 + (lane >> 4) * 4 + r;
      const int row_g = t0 + row_l;
      if (row_g < T) {
        const float a = half_lse[row_l];
        const float b = half_lse[64 + row_l];
        const float mx = fmaxf(a, b);
        lse_out[row_g] = mx + __logf(
            __builtin_amdgcn_exp2f((a - mx) * kLog2eC) +
            __builtin_amdgcn_exp2f((b - mx) * kLog2eC));
        tgt_out[row_g] = fmaxf(half_tgt[row_l], half_tgt[64 + row_l]);
      }
    }
  }
}

}  // namespace d9d

std::vector<torch::Tensor> cce_fwd(
    torch::Tensor e, torch::Tensor c, torch::Tensor targets) {
  TORCH_CHECK(e.is_cuda() && e.scalar_type() == torch::kBFloat16 && e.is_contiguous());
  TORCH_CHECK(c.scalar_type() == torch::kBFloat16 && c.is_contiguous());
  TORCH_CHECK(targets.scalar_type() == torch::kInt64);
  const int T = e.size(0), K = e.size(1), V = c.size(0);
  TORCH_CHECK(K % 64 == 0, "cce_fwd requires hidden % 64 == 0");
  const bool wide = (K % 128 == 0) &&
      (size_t)(64 * K + 128 * 128) * sizeof(__bf16) <= 160 * 1024;
  const size_t smem =
      (size_t)(64 * K + 128 * (wide ? 128 : 64)) * sizeof(__bf16);
  TORCH_CHECK(smem <= 160 * 1024, "cce_fwd: hidden too large for LDS: ", K);

  auto lse = torch::empty({T}, e.options().dtype(torch::kFloat32));
  auto tgt = torch::empty({T}, e.options().dtype(torch::kFloat32));
  if (T == 0) return {lse, tgt};
  auto stream = at::hip::getCurrentHIPStream();
  auto tgts = targets.contiguous();
#define LAUNCH_CCE(KT)                                                        \
  hipLaunchKernelGGL(d9d::cce_fwd_kernel<KT>, dim3((T + 63) / 64), dim3(512), \
                     smem, stream,                                            \
                     reinterpret_cast<const __bf16*>(e.data_ptr()),           \
                     reinterpret_cast<const __bf16*>(c.data_ptr()),           \
                     tgts.data_ptr<int64_t>(), lse.data_ptr<float>(),         \
                     tgt.data_ptr<float>(), T, V, K)
  if (wide) LAUNCH_CCE(128); else LAUNCH_CCE(64);
#undef LAUNCH_CCE
  return {lse, tgt};
}

namespace d9d {

// In-place CCE backward dlogits: p = exp(logit - lse) (fp32 math), minus the
// one-hot target, scaled by the per-row upstream grad -- replaces a 4-pass
// torch chain (sub, exp_, scatter_add_, mul_) over the (Tc, V) chunk with
// one read+write.
__global__ void cce_dlogits_kernel(
    ushort* __restrict__ logits,       // (R, V) bf16, in/out
    const float* __restrict__ lse,     // (R,)
    const int64_t* __restrict__ targets,  // (R,) GLOBAL vocab ids (or ignore)
    const float* __restrict__ dl,      // (R,) upstream grad (0 for ignored)
    const float* __restrict__ dlse,    // (R,) upstream lse grad, or nullptr
    int64_t R, int64_t V, int64_t vocab_start, int64_t ignore_index,
    float filter_eps) {  // zero non-target dlogits with p < eps (<=0: off)
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t row = blockIdx.y; row < R; row += gridDim.y) {
  const float l = lse[row];
  const float g = dl[row];
  // d(lse)/d(logit) = p: a differentiable lse output simply adds its grad
  // to the softmax term's scale (loss term keeps -onehot * g).
  const float gp = g + (dlse ? dlse[row] : 0.f);
  const int64_t tgt_global = targets[row];
  const int64_t tgt =
      (tgt_global == ignore_index) ? -1 : tgt_global - vocab_start;
  ushort* rowp = logits + row * V;

  for (int64_t base = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       base < V; base += stride) {
    if (base + 8 <= V) {
      Bf16x8 x;
      x.u = *reinterpret_cast<const ushort8v*>(rowp + base);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = __builtin_amdgcn_exp2f(
            (bf16_bits_to_f32(x.s[j]) - l) * 1.44269504089f);
        const bool is_tgt = base + j == tgt;
        if (p < filter_eps && !is_tgt) p = 0.f;
        float out = p * gp - (is_tgt ? g : 0.f);
        x.s[j] = f32_to_bf16_rne(out);
      }
      *reinterpret_cast<ushort8v*>(rowp + base) = x.u;
    } else {
      for (int64_t i = base; i < V; ++i) {
        float p = __builtin_amdgcn_exp2f(
            (bf16_bits_to_f32(rowp[i]) - l) * 1.44269504089f);
        const bool is_tgt = i == tgt;
        if (p < filter_eps && !is_tgt) p = 0.f;
        float out = p * gp - (is_tgt ? g : 0.f);
        rowp[i] = f32_to_bf16_rne(out);
      }
    }
  }
  }
}

}  // namespace d9d

torch::Tensor cce_dlogits_(torch::Tensor logits, torch::Tensor lse,
                           torch::Tensor targets, torch::Tensor dl,
                           c10::optional<torch::Tensor> dlse,
                           int64_t vocab_start, int64_t ignore_index,
                           double filter_eps) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == torch::kBFloat16 &&
              logits.is_contiguous());
  const int64_t R = logits.size(0), V = logits.size(1);
  if (R == 0) return logits;
  const int gx = (int)std::min<int64_t>((V + 256 * 8 - 1) / (256 * 8), 128);
  const unsigned gy = (unsigned)std::min<int64_t>(R, 4096 / gx + 1);
  auto stream = at::hip::getCurrentHIPStream();
  torch::Tensor dlse_c;
  const float* dlse_ptr = nullptr;
  if (dlse.has_value()) {
    dlse_c = dlse->contiguous();
    dlse_ptr = dlse_c.data_ptr<float>();
  }
  hipLaunchKernelGGL(d9d::cce_dlogits_kernel, dim3(gx, gy), dim3(256),
                     0, stream,
                     reinterpret_cast<ushort*>(logits.data_ptr()),
                     lse.contiguous().data_ptr<float>(),
                     targets.contiguous().data_ptr<int64_t>(),
                     dl.contiguous().data_ptr<float>(), dlse_ptr,
                     R, V, vocab_start,
                     ignore_index, (float)filter_eps);
  return logits;
}
