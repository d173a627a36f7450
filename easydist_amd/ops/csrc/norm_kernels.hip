// LayerNorm / RMSNorm forward+backward for gfx950.
//
// Memory-bound ops: the design target is the HBM3E roofline (~6.3 TB/s
// achievable). One WAVE per row, bf16x8 / float4 vectorized loads
// (16 B/lane), fp32 accumulation, wave shuffle reductions; dweight/dbias
// accumulate into fp32 global scratch with device-scope atomics.
// Replaces aten.native_layer_norm(+backward) via the lower_hip pass.
#include "common.h"
#include <string>
#include <cstdlib>

// ---------------------------------------------------------------- fwd -------
template <typename T, bool RMS>
__global__ void norm_fwd_kernel(const T* __restrict__ x,
                                const T* __restrict__ w,
                                const T* __restrict__ b,
                                T* __restrict__ out,
                                float* __restrict__ mean_out,
                                float* __restrict__ rstd_out,
                                int rows, int D, float eps) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (wave_id >= rows) return;
  const T* xrow = x + (long)wave_id * D;
  T* orow = out + (long)wave_id * D;

  const int VEC = 8;
  float s1 = 0.f, s2 = 0.f;
  for (int i = lane * VEC; i < D; i += WAVE * VEC) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(&xrow[i]);
    #pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float f = bf2f(xv.v[j]);
      s1 += f;
      s2 += f * f;
    }
  }
  s1 = wave_sum(s1);
  s2 = wave_sum(s2);
  float mean, rstd;
  if (RMS) {
    mean = 0.f;
    rstd = rsqrtf(s2 / D + eps);
  } else {
    mean = s1 / D;
    rstd = rsqrtf(fmaxf(s2 / D - mean * mean, 0.f) + eps);
  }
  if (lane == 0) {
    if (!RMS) mean_out[wave_id] = mean;
    rstd_out[wave_id] = rstd;
  }
  for (int i = lane * VEC; i < D; i += WAVE * VEC) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(&xrow[i]);
    bf16x8 ov;
    bf16x8 wv, bv;
    if (w) wv = *reinterpret_cast<const bf16x8*>(&w[i]);
    if (b) bv = *reinterpret_cast<const bf16x8*>(&b[i]);
    #pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float f = (bf2f(xv.v[j]) - mean) * rstd;
      if (w) f *= bf2f(wv.v[j]);
      if (b) f += bf2f(bv.v[j]);
      ov.v[j] = f2bf(f);
    }
    *reinterpret_cast<bf16x8*>(&orow[i]) = ov;
  }
}

// fp32 variant
template <bool RMS>
__global__ void norm_fwd_kernel_f32(const float* __restrict__ x,
                                    const float* __restrict__ w,
                                    const float* __restrict__ b,
                                    float* __restrict__ out,
                                    float* __restrict__ mean_out,
                                    float* __restrict__ rstd_out,
                                    int rows, int D, float eps) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (wave_id >= rows) return;
  const float* xrow = x + (long)wave_id * D;
  float* orow = out + (long)wave_id * D;
  const int VEC = 4;
  float s1 = 0.f, s2 = 0.f;
  for (int i = lane * VEC; i < D; i += WAVE * VEC) {
    float4 xv = *reinterpret_cast<const float4*>(&xrow[i]);
    s1 += xv.x + xv.y + xv.z + xv.w;
    s2 += xv.x * xv.x + xv.y * xv.y + xv.z * xv.z + xv.w * xv.w;
  }
  s1 = wave_sum(s1);
  s2 = wave_sum(s2);
  float mean, rstd;
  if (RMS) {
    mean = 0.f;
    rstd = rsqrtf(s2 / D + eps);
  } else {
    mean = s1 / D;
    rstd = rsqrtf(fmaxf(s2 / D - mean * mean, 0.f) + eps);
  }
  if (lane == 0) {
    if (!RMS) mean_out[wave_id] = mean;
    rstd_out[wave_id] = rstd;
  }
  for (int i = lane * VEC; i < D; i += WAVE * VEC) {
    float4 xv = *reinterpret_cast<const float4*>(&xrow[i]);
    float4 ov;
    float* xp = &xv.x;
    float* op = &ov.x;
    #pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float f = (xp[j] - mean) * rstd;
      if (w) f *= w[i + j];
      if (b) f += b[i + j];
      op[j] = f;
    }
    *reinterpret_cast<float4*>(&orow[i]) = ov;
  }
}

// ---------------------------------------------------------------- bwd -------
// dx_j = rstd * ( gw_j - mean(gw) - xhat_j * mean(gw * xhat) )   [LN]
// where gw = grad * w. dw_j = sum_rows grad_j * xhat_j ; db_j = sum_rows grad.
template <typename T, bool RMS>
__global__ void norm_bwd_kernel(const T* __restrict__ grad,
                                const T* __restrict__ x,
                                const float* __restrict__ mean,
                                const float* __restrict__ rstd,
                                const T* __restrict__ w,
                                T* __restrict__ dx,
                                float* __restrict__ dw,   // fp32 scratch [D]
                                float* __restrict__ db,   // fp32 scratch [D]
                                int rows, int D) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (wave_id >= rows) return;
  const T* grow = grad + (long)wave_id * D;
  const T* xrow = x + (long)wave_id * D;
  T* dxrow = dx + (long)wave_id * D;
  const float m = RMS ? 0.f : mean[wave_id];
  const float r = rstd[wave_id];

  const int VEC = 8;
  float sum_gw = 0.f, sum_gwx = 0.f;
  for (int i = lane * VEC; i < D; i += WAVE * VEC) {
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(&grow[i]);
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(&xrow[i]);
    bf16x8 wv;
    if (w) wv = *reinterpret_cast<const bf16x8*>(&w[i]);
    #pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float g = bf2f(gv.v[j]);
      float xh = (bf2f(xv.v[j]) - m) * r;
      float gw = w ? g * bf2f(wv.v[j]) : g;
      sum_gw += gw;
      sum_gwx += gw * xh;
    }
  }
  sum_gw = wave_sum(sum_gw) / D;
  sum_gwx = wave_sum(sum_gwx) / D;
  for (int i = lane * VEC; i < D; i += WAVE * VEC) {
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(&grow[i]);
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(&xrow[i]);
    bf16x8 wv;
    if (w) wv = *reinterpret_cast<const bf16x8*>(&w[i]);
    bf16x8 dxv;
    #pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float g = bf2f(gv.v[j]);
      float xh = (bf2f(xv.v[j]) - m) * r;
      float gw = w ? g * bf2f(wv.v[j]) : g;
      float v = RMS ? r * (gw - xh * sum_gwx)
                    : r * (gw - sum_gw - xh * sum_gwx);
      dxv.v[j] = f2bf(v);
    }
    *reinterpret_cast<bf16x8*>(&dxrow[i]) = dxv;
  }
}

// Column reduction for dweight/dbias: each block owns 256 columns x a chunk
// of rows, accumulates in registers, ONE atomic per (block, column).
template <typename T, bool RMS>
__global__ void norm_param_grads_kernel(const T* __restrict__ grad,
                                        const T* __restrict__ x,
                                        const float* __restrict__ mean,
                                        const float* __restrict__ rstd,
                                        float* __restrict__ dw,
                                        float* __restrict__ db,
                                        int rows, int D, int rows_per_blk) {
  // 8 columns per thread with 16-B vector loads: the scalar-bf16 form
  // streamed at half rate (guide common-mistake 2)
  const int col0 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (col0 >= D) return;
  const int r0 = blockIdx.y * rows_per_blk;
  const int r1 = min(r0 + rows_per_blk, rows);
  float sw[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float sb[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int r = r0; r < r1; ++r) {
    const bf16x8 g8 = *reinterpret_cast<const bf16x8*>(
        &grad[(long)r * D + col0]);
    const bf16x8 x8 = *reinterpret_cast<const bf16x8*>(
        &x[(long)r * D + col0]);
    const float m = RMS ? 0.f : mean[r];
    const float rs = rstd[r];
    #pragma unroll
    for (int u = 0; u < 8; ++u) {
      float g = bf2f(g8.v[u]);
      sw[u] += g * (bf2f(x8.v[u]) - m) * rs;
      sb[u] += g;
    }
  }
  #pragma unroll
  for (int u = 0; u < 8; ++u) {
    atomicAdd(&dw[col0 + u], sw[u]);
    atomicAdd(&db[col0 + u], sb[u]);
  }
}

template <typename T, bool RMS>
__global__ void norm_param_grads_scalar_kernel(const T* __restrict__ grad,
                                               const T* __restrict__ x,
                                               const float* __restrict__ mean,
                                               const float* __restrict__ rstd,
                                               float* __restrict__ dw,
                                               float* __restrict__ db,
                                               int rows, int D,
                                               int rows_per_blk) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= D) return;
  const int r0 = blockIdx.y * rows_per_blk;
  const int r1 = min(r0 + rows_per_blk, rows);
  float sw = 0.f, sb = 0.f;
  // unroll keeps ~8 independent load pairs in flight per lane — the
  // rolled form was latency-bound at half of HBM bandwidth
  #pragma unroll 8
  for (int r = r0; r < r1; ++r) {
    float g = bf2f(grad[(long)r * D + col]);
    float m = RMS ? 0.f : mean[r];
    float xh = (bf2f(x[(long)r * D + col]) - m) * rstd[r];
    sw += g * xh;
    sb += g;
  }
  atomicAdd(&dw[col], sw);
  atomicAdd(&db[col], sb);
}

static bool lnpg_scalar() {
  // measured on MI355X (65536x768): scalar-column form 0.166 ms/call vs
  // 0.221 for the vectorized+fewer-atomics form — per-address atomic
  // count and full-block occupancy beat wider loads here
  static int v = [] {
    const char* e = getenv("EASYDIST_LNPG");
    return (e && std::string(e) == "vector") ? 0 : 1;
  }();
  return v;
}

// ---------------------------------------------------------- host wrappers ---
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

static void check_lastdim(const at::Tensor& x) {
  TORCH_CHECK(x.is_contiguous(), "norm kernels need contiguous input");
  TORCH_CHECK(x.size(-1) % 8 == 0, "feature dim must be a multiple of 8");
}

static void check_affine(const at::Tensor& x,
                         const std::optional<at::Tensor>& w,
                         const std::optional<at::Tensor>& b) {
  if (w) TORCH_CHECK(w->numel() == x.size(-1), "weight/feature mismatch");
  if (b) TORCH_CHECK(b->numel() == x.size(-1), "bias/feature mismatch");
}

std::tuple<at::Tensor, at::Tensor, at::Tensor>
layer_norm_fwd(const at::Tensor& x, const std::optional<at::Tensor>& w,
               const std::optional<at::Tensor>& b, double eps) {
  check_lastdim(x);
  check_affine(x, w, b);
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto out = at::empty_like(x);
  auto stat_shape = x.sizes().vec();
  stat_shape.back() = 1;
  auto mean = at::empty(stat_shape, x.options().dtype(at::kFloat));
  auto rstd = at::empty(stat_shape, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const int WPB = 4;   // waves per block
  dim3 block(WAVE * WPB), grid((rows + WPB - 1) / WPB);
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((norm_fwd_kernel<bf16, false>), grid, block, 0, stream,
        (const bf16*)x.data_ptr(), w ? (const bf16*)w->data_ptr() : nullptr,
        b ? (const bf16*)b->data_ptr() : nullptr, (bf16*)out.data_ptr(),
        mean.data_ptr<float>(), rstd.data_ptr<float>(), rows, D, (float)eps);
  } else {
    hipLaunchKernelGGL((norm_fwd_kernel_f32<false>), grid, block, 0, stream,
        x.data_ptr<float>(), w ? w->data_ptr<float>() : nullptr,
        b ? b->data_ptr<float>() : nullptr, out.data_ptr<float>(),
        mean.data_ptr<float>(), rstd.data_ptr<float>(), rows, D, (float)eps);
  }
  return {out, mean, rstd};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor>
layer_norm_bwd(const at::Tensor& grad, const at::Tensor& x,
               const at::Tensor& mean, const at::Tensor& rstd,
               const std::optional<at::Tensor>& w,
               std::vector<bool> mask) {
  check_lastdim(x);
  check_affine(x, w, std::nullopt);
  TORCH_CHECK(grad.sizes() == x.sizes() && grad.is_contiguous());
  TORCH_CHECK(mean.numel() == x.numel() / x.size(-1));
  TORCH_CHECK(x.scalar_type() == at::kBFloat16,
              "layer_norm_bwd kernel: bf16 only (fp32 falls back to aten)");
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto dx = at::empty_like(x);
  auto dwf = at::zeros({D}, x.options().dtype(at::kFloat));
  auto dbf = at::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const int WPB = 4;
  dim3 block(WAVE * WPB), grid((rows + WPB - 1) / WPB);
  hipLaunchKernelGGL((norm_bwd_kernel<bf16, false>), grid, block, 0, stream,
      (const bf16*)grad.data_ptr(), (const bf16*)x.data_ptr(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(),
      w ? (const bf16*)w->data_ptr() : nullptr, (bf16*)dx.data_ptr(),
      dwf.data_ptr<float>(), dbf.data_ptr<float>(), rows, D);
  // one-wave blocks, 512 columns each (64 lanes x 8 cols of 16-B
  // loads); y splits rows so x*y fills the 256 CUs
  const int gx = (D + 511) / 512;
  // few row-chunks: every block atomicAdds the whole 2*D output, so
  // block count IS the atomic contention per address
  int rows_per_blk = 256;
  while ((long)gx * ((rows + rows_per_blk - 1) / rows_per_blk) > 1024)
    rows_per_blk *= 2;
  dim3 gblock(64), ggrid(gx, (rows + rows_per_blk - 1) / rows_per_blk);
  if (lnpg_scalar()) {
    const int rpb2 = 256;
    dim3 sb2(256), sg2((D + 255) / 256, (rows + rpb2 - 1) / rpb2);
    hipLaunchKernelGGL((norm_param_grads_scalar_kernel<bf16, false>), sg2,
        sb2, 0, stream, (const bf16*)grad.data_ptr(),
        (const bf16*)x.data_ptr(), mean.data_ptr<float>(),
        rstd.data_ptr<float>(), dwf.data_ptr<float>(),
        dbf.data_ptr<float>(), rows, D, rpb2);
    auto dtype0 = at::kFloat;
    return {dx, dwf.to(dtype0), dbf.to(dtype0)};
  }
  hipLaunchKernelGGL((norm_param_grads_kernel<bf16, false>), ggrid, gblock, 0,
      stream, (const bf16*)grad.data_ptr(), (const bf16*)x.data_ptr(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(), dwf.data_ptr<float>(),
      dbf.data_ptr<float>(), rows, D, rows_per_blk);
  auto dtype = at::kFloat;   // grads feed fp32 Adam: never truncate
  return {dx, dwf.to(dtype), dbf.to(dtype)};
}

std::tuple<at::Tensor, at::Tensor>
rms_norm_fwd(const at::Tensor& x, const std::optional<at::Tensor>& w,
             double eps) {
  check_lastdim(x);
  check_affine(x, w, std::nullopt);
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto out = at::empty_like(x);
  auto stat_shape = x.sizes().vec();
  stat_shape.back() = 1;
  auto rstd = at::empty(stat_shape, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const int WPB = 4;
  dim3 block(WAVE * WPB), grid((rows + WPB - 1) / WPB);
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((norm_fwd_kernel<bf16, true>), grid, block, 0, stream,
        (const bf16*)x.data_ptr(), w ? (const bf16*)w->data_ptr() : nullptr,
        nullptr, (bf16*)out.data_ptr(), nullptr, rstd.data_ptr<float>(),
        rows, D, (float)eps);
  } else {
    hipLaunchKernelGGL((norm_fwd_kernel_f32<true>), grid, block, 0, stream,
        x.data_ptr<float>(), w ? w->data_ptr<float>() : nullptr, nullptr,
        out.data_ptr<float>(), nullptr, rstd.data_ptr<float>(), rows, D,
        (float)eps);
  }
  return {out, rstd};
}

std::tuple<at::Tensor, at::Tensor>
rms_norm_bwd(const at::Tensor& grad, const at::Tensor& x,
             const at::Tensor& rstd, const std::optional<at::Tensor>& w) {
  check_lastdim(x);
  check_affine(x, w, std::nullopt);
  TORCH_CHECK(grad.sizes() == x.sizes() && grad.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "rms_norm_bwd: bf16 only");
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto dx = at::empty_like(x);
  auto dwf = at::zeros({D}, x.options().dtype(at::kFloat));
  auto dbf = at::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const int WPB = 4;
  dim3 block(WAVE * WPB), grid((rows + WPB - 1) / WPB);
  hipLaunchKernelGGL((norm_bwd_kernel<bf16, true>), grid, block, 0, stream,
      (const bf16*)grad.data_ptr(), (const bf16*)x.data_ptr(), nullptr,
      rstd.data_ptr<float>(), w ? (const bf16*)w->data_ptr() : nullptr,
      (bf16*)dx.data_ptr(), dwf.data_ptr<float>(), dbf.data_ptr<float>(),
      rows, D);
  // one-wave blocks, 512 columns each (64 lanes x 8 cols of 16-B
  // loads); y splits rows so x*y fills the 256 CUs
  const int gx = (D + 511) / 512;
  // few row-chunks: every block atomicAdds the whole 2*D output, so
  // block count IS the atomic contention per address
  int rows_per_blk = 256;
  while ((long)gx * ((rows + rows_per_blk - 1) / rows_per_blk) > 1024)
    rows_per_blk *= 2;
  dim3 gblock(64), ggrid(gx, (rows + rows_per_blk - 1) / rows_per_blk);
  hipLaunchKernelGGL((norm_param_grads_kernel<bf16, true>), ggrid, gblock, 0,
      stream, (const bf16*)grad.data_ptr(), (const bf16*)x.data_ptr(),
      nullptr, rstd.data_ptr<float>(), dwf.data_ptr<float>(),
      dbf.data_ptr<float>(), rows, D, rows_per_blk);
  auto dtype = at::kFloat;   // grads feed fp32 Adam: never truncate
  return {dx, dwf.to(dtype)};
}
