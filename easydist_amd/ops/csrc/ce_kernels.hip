// Fused cross-entropy over a large vocab for gfx950.
//
// fwd: one wave per row streams the logits once (bf16x8 / float4 loads),
// computing max and sum-exp online (no [N,V] log-softmax materialized —
// saves ~1.6 GB fp32 at GPT-2 shapes). bwd writes softmax-minus-onehot
// scaled by grad/N in one pass.
#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

template <typename T>
DEVINL float loadf(const T* p, int i);
template <> DEVINL float loadf<bf16>(const bf16* p, int i) { return bf2f(p[i]); }
template <> DEVINL float loadf<float>(const float* p, int i) { return p[i]; }

// 8-element vector load/store (one 16B dword4 for bf16, two for fp32)
DEVINL void vload8(const bf16* p, int i, float* out) {
  bf16x8 v = *reinterpret_cast<const bf16x8*>(&p[i]);
  #pragma unroll
  for (int u = 0; u < 8; ++u) out[u] = bf2f(v.v[u]);
}
DEVINL void vload8(const float* p, int i, float* out) {
  f32x4v a = *reinterpret_cast<const f32x4v*>(&p[i]);
  f32x4v b = *reinterpret_cast<const f32x4v*>(&p[i + 4]);
  out[0] = a.x; out[1] = a.y; out[2] = a.z; out[3] = a.w;
  out[4] = b.x; out[5] = b.y; out[6] = b.z; out[7] = b.w;
}
DEVINL void vstore8(bf16* p, int i, const bf16* v) {
  bf16x8 o;
  #pragma unroll
  for (int u = 0; u < 8; ++u) o.v[u] = v[u];
  *reinterpret_cast<bf16x8*>(&p[i]) = o;
}
DEVINL void vstore8(float* p, int i, const float* v) {
  #pragma unroll
  for (int u = 0; u < 8; ++u) p[i + u] = v[u];
}

template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ nll,
                              float* __restrict__ lse_out,
                              int N, int V) {
  const int row = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (row >= N) return;
  const T* lrow = logits + (long)row * V;
  // online max + sumexp; 8-wide vector loads (16B bf16 / 32B fp32 per
  // lane-iteration) keep the 8 TB/s HBM stream coalesced
  float m = -INFINITY, s = 0.f;
  const int V8 = (V / 8) * 8;
  for (int i = lane * 8; i < V8; i += WAVE * 8) {
    float v8[8];
    vload8(lrow, i, v8);
    #pragma unroll
    for (int u = 0; u < 8; ++u) {
      float m2 = fmaxf(m, v8[u]);
      s = s * __expf(m - m2) + __expf(v8[u] - m2);
      m = m2;
    }
  }
  for (int i = V8 + lane; i < V; i += WAVE) {
    float v = loadf(lrow, i);
    float m2 = fmaxf(m, v);
    s = s * __expf(m - m2) + __expf(v - m2);
    m = m2;
  }
  // combine lanes
  float gm = wave_max(m);
  s = s * __expf(m - gm);
  s = wave_sum(s);
  float lse = gm + __logf(s);
  if (lane == 0) {
    lse_out[row] = lse;
    nll[row] = lse - loadf(lrow, (int)targets[row]);
  }
}

template <typename T, bool PER_ROW>
__global__ void ce_bwd_kernel(const float* __restrict__ grad_scalar,
                              const T* __restrict__ logits,
                              const long* __restrict__ targets,
                              const float* __restrict__ lse,
                              T* __restrict__ dlogits,
                              int N, int V) {
  const int row = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (row >= N) return;
  const T* lrow = logits + (long)row * V;
  T* drow = dlogits + (long)row * V;
  // SUM semantics (scalar) or per-row grad (the lowered nll_loss path)
  const float scale = PER_ROW ? grad_scalar[row] : grad_scalar[0];
  const float l = lse[row];
  const long tg = targets[row];
  const int V8 = (V / 8) * 8;
  for (int i = lane * 8; i < V8; i += WAVE * 8) {
    float v8[8];
    vload8(lrow, i, v8);
    T out8[8];
    #pragma unroll
    for (int u = 0; u < 8; ++u) {
      float p = __expf(v8[u] - l);
      if (i + u == tg) p -= 1.f;
      if constexpr (sizeof(T) == 2) out8[u] = f2bf(p * scale);
      else out8[u] = p * scale;
    }
    vstore8(drow, i, out8);
  }
  for (int i = V8 + lane; i < V; i += WAVE) {
    float p = __expf(loadf(lrow, i) - l);
    if (i == tg) p -= 1.f;
    if constexpr (sizeof(T) == 2) drow[i] = f2bf(p * scale);
    else drow[i] = p * scale;
  }
}

std::tuple<at::Tensor, at::Tensor> ce_fwd_rows(const at::Tensor& logits,
                                               const at::Tensor& targets) {
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(targets.scalar_type() == at::kLong);
  TORCH_CHECK(targets.numel() == logits.size(0), "targets/rows mismatch");
  const int N = logits.size(0), V = logits.size(1);
  auto nll = at::empty({N}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({N}, logits.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const int WPB = 4;
  dim3 block(WAVE * WPB), grid((N + WPB - 1) / WPB);
  if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(ce_fwd_kernel<bf16>, grid, block, 0, stream,
        (const bf16*)logits.data_ptr(), targets.data_ptr<long>(),
        nll.data_ptr<float>(), lse.data_ptr<float>(), N, V);
  } else {
    hipLaunchKernelGGL(ce_fwd_kernel<float>, grid, block, 0, stream,
        logits.data_ptr<float>(), targets.data_ptr<long>(),
        nll.data_ptr<float>(), lse.data_ptr<float>(), N, V);
  }
  return {nll, lse};
}

std::tuple<at::Tensor, at::Tensor> ce_fwd(const at::Tensor& logits,
                                          const at::Tensor& targets) {
  auto [nll, lse] = ce_fwd_rows(logits, targets);
  return {nll.sum(), lse};
}

at::Tensor ce_bwd(const at::Tensor& grad, const at::Tensor& logits,
                  const at::Tensor& targets, const at::Tensor& lse) {
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(targets.numel() == logits.size(0));
  TORCH_CHECK(lse.numel() == logits.size(0));
  const int N = logits.size(0), V = logits.size(1);
  auto dlogits = at::empty_like(logits);
  auto gradf = grad.to(at::kFloat).contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  const int WPB = 4;
  dim3 block(WAVE * WPB), grid((N + WPB - 1) / WPB);
  const bool per_row = gradf.numel() == N;
  TORCH_CHECK(per_row || gradf.numel() == 1, "ce_bwd: grad scalar or [N]");
  if (logits.scalar_type() == at::kBFloat16) {
    auto kern = per_row ? ce_bwd_kernel<bf16, true>
                        : ce_bwd_kernel<bf16, false>;
    hipLaunchKernelGGL(kern, grid, block, 0, stream,
        gradf.data_ptr<float>(), (const bf16*)logits.data_ptr(),
        targets.data_ptr<long>(), lse.data_ptr<float>(),
        (bf16*)dlogits.data_ptr(), N, V);
  } else {
    auto kern = per_row ? ce_bwd_kernel<float, true>
                        : ce_bwd_kernel<float, false>;
    hipLaunchKernelGGL(kern, grid, block, 0, stream,
        gradf.data_ptr<float>(), logits.data_ptr<float>(),
        targets.data_ptr<long>(), lse.data_ptr<float>(),
        dlogits.data_ptr<float>(), N, V);
  }
  return dlogits;
}
