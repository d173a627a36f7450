// Fused multi-tensor Adam step for gfx950.
//
// One grid-stride kernel walks a chunk table covering every parameter
// shard: the per-step optimizer tail collapses from ~4 launches per
// parameter to ONE launch. fp32 math; functional form (writes fresh
// outputs) so the pure sharded graph + hipGraph capture stay clean.
// The chunk table is cached on device keyed by the input pointer set, so
// replays (hipGraph) never rebuild or re-upload it.
#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <ATen/cuda/CUDAGraphsUtils.cuh>
#include <map>
#include <utility>
#include <vector>

struct AdamChunk {
  const float* p;
  const void* g;        // fp32 or bf16 (g_bf16 flag of the launch)
  const float* ea;
  const float* eas;
  float* out_p;
  float* out_ea;
  float* out_eas;
  int tensor_idx;
  int n;          // elements in this chunk
};

DEVINL void load_g4(const void* g, int i, int g_bf16, float* out) {
  if (g_bf16) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(&((const bf16*)g)[i & ~7]);
    const int o = i & 4;
    #pragma unroll
    for (int j = 0; j < 4; ++j) out[j] = bf2f(v.v[o + j]);
  } else {
    float4 v = *reinterpret_cast<const float4*>(&((const float*)g)[i]);
    out[0] = v.x; out[1] = v.y; out[2] = v.z; out[3] = v.w;
  }
}

#define CHUNK_ELEMS 65536

// ---------------------------------------------------------------------------
// capture-safe chunk-table staging, keyed by (table bytes, first param
// pointer): the optimizer fuse pass emits SEPARATE fused steps per grad
// dtype bank, so ONE static pair is not enough — during hipGraph capture
// the first call retires its recorded buffers and the second call must
// find its OWN warm pair (sized by the eager warmup, same key).
// ---------------------------------------------------------------------------
struct StagingPair {
  at::Tensor pinned, dev;
  size_t capacity = 0;
  hipEvent_t h2d_done = nullptr;
};

static StagingPair& staging_for(std::map<std::pair<size_t, const void*>,
                                         StagingPair>& pool,
                                size_t bytes, const void* key_ptr,
                                const at::Device& dev, bool capturing,
                                const char* who) {
  auto& st = pool[{bytes, key_ptr}];
  if (bytes > st.capacity) {
    TORCH_CHECK(!capturing, who,
                ": staging buffers must be warmed up (one eager call per "
                "parameter bank) before hipGraph capture");
    st.pinned = at::empty({(long)bytes},
                          at::TensorOptions().dtype(at::kByte)
                              .pinned_memory(true));
    st.dev = at::empty({(long)bytes},
                       at::TensorOptions().dtype(at::kByte).device(dev));
    st.capacity = bytes;
  }
  if (st.h2d_done == nullptr)
    C10_CUDA_CHECK(hipEventCreateWithFlags(&st.h2d_done,
                                           hipEventDisableTiming));
  else if (!capturing)
    // the PREVIOUS call's async H2D must have consumed the pinned buffer
    // before the host overwrites it. hipEventSynchronize is REJECTED
    // inside capture (the runtime device-synchronizes right before
    // capture begins, which retires any in-flight eager H2D).
    C10_CUDA_CHECK(hipEventSynchronize(st.h2d_done));
  return st;
}

static void staging_finish(std::map<std::pair<size_t, const void*>,
                                    StagingPair>& pool,
                           size_t bytes, const void* key_ptr,
                           bool capturing, hipStream_t stream) {
  auto it = pool.find({bytes, key_ptr});
  if (it == pool.end()) return;
  if (capturing) {
    // the recorded H2D re-reads THIS pinned buffer on every replay:
    // retire the pair so no later call can overwrite it
    static std::vector<at::Tensor> keepalive;
    keepalive.push_back(it->second.pinned);
    keepalive.push_back(it->second.dev);
    pool.erase(it);
  } else {
    C10_CUDA_CHECK(hipEventRecord(it->second.h2d_done, stream));
  }
}

__global__ void fused_adam_kernel(const AdamChunk* __restrict__ chunks,
                                  int n_chunks, int g_bf16,
                                  const float* __restrict__ steps,  // [T]
                                  float lr, float beta1, float beta2,
                                  float weight_decay, float eps) {
  const int c = blockIdx.x;
  if (c >= n_chunks) return;
  const AdamChunk ch = chunks[c];
  const float step = steps[ch.tensor_idx] + 1.f;
  const float bc1 = 1.f - __powf(beta1, step);
  const float bc2 = 1.f - __powf(beta2, step);
  const float inv_bc1 = 1.f / bc1;
  const float rsqrt_bc2 = rsqrtf(bc2);
  // unroll keeps several independent 16-B load groups in flight per lane
  #pragma unroll 4
  for (int i = threadIdx.x * 4; i + 3 < ch.n; i += blockDim.x * 4) {
    float4 p = *reinterpret_cast<const float4*>(&ch.p[i]);
    float gg[4];
    load_g4(ch.g, i, g_bf16, gg);
    float4 ea = *reinterpret_cast<const float4*>(&ch.ea[i]);
    float4 eas = *reinterpret_cast<const float4*>(&ch.eas[i]);
    float* pp = &p.x; float* ee = &ea.x; float* ss = &eas.x;
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float grad = gg[j];
      if (weight_decay != 0.f) grad += weight_decay * pp[j];
      float m = beta1 * ee[j] + (1.f - beta1) * grad;
      float v = beta2 * ss[j] + (1.f - beta2) * grad * grad;
      float denom = __fsqrt_rn(v) * rsqrt_bc2 + eps;
      pp[j] = pp[j] - lr * (m * inv_bc1) / denom;
      ee[j] = m;
      ss[j] = v;
    }
    *reinterpret_cast<float4*>(&ch.out_p[i]) = p;
    *reinterpret_cast<float4*>(&ch.out_ea[i]) = ea;
    *reinterpret_cast<float4*>(&ch.out_eas[i]) = eas;
  }
  // scalar tail
  int tail_start = (ch.n / 4) * 4;
  for (int i = tail_start + threadIdx.x; i < ch.n; i += blockDim.x) {
    float grad = g_bf16 ? bf2f(((const bf16*)ch.g)[i])
                        : ((const float*)ch.g)[i];
    if (weight_decay != 0.f) grad += weight_decay * ch.p[i];
    float m = beta1 * ch.ea[i] + (1.f - beta1) * grad;
    float v = beta2 * ch.eas[i] + (1.f - beta2) * grad * grad;
    float denom = __fsqrt_rn(v) * rsqrt_bc2 + eps;
    ch.out_p[i] = ch.p[i] - lr * (m * inv_bc1) / denom;
    ch.out_ea[i] = m;
    ch.out_eas[i] = v;
  }
}

__global__ void bump_steps_kernel(const float* const* __restrict__ in_steps,
                                  float* const* __restrict__ out_steps,
                                  int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out_steps[i][0] = in_steps[i][0] + 1.f;
}

std::tuple<std::vector<at::Tensor>, std::vector<at::Tensor>,
           std::vector<at::Tensor>, std::vector<at::Tensor>>
fused_adam_step(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                std::vector<at::Tensor> exp_avgs,
                std::vector<at::Tensor> exp_avg_sqs,
                std::vector<at::Tensor> steps, double lr, double beta1,
                double beta2, double weight_decay, double eps) {
  const int T = params.size();
  std::vector<at::Tensor> out_p, out_ea, out_eas, out_steps;
  out_p.reserve(T); out_ea.reserve(T); out_eas.reserve(T);
  const bool g_bf16 = !grads.empty()
      && grads[0].scalar_type() == at::kBFloat16;
  for (int t = 0; t < T; ++t) {
    TORCH_CHECK(params[t].scalar_type() == at::kFloat, "fp32 params only");
    TORCH_CHECK(grads[t].scalar_type()
                    == (g_bf16 ? at::kBFloat16 : at::kFloat),
                "grads must be uniformly fp32 or bf16");
    TORCH_CHECK(params[t].is_contiguous() && grads[t].is_contiguous());
    out_p.push_back(at::empty_like(params[t]));
    out_ea.push_back(at::empty_like(exp_avgs[t]));
    out_eas.push_back(at::empty_like(exp_avg_sqs[t]));
    out_steps.push_back(at::empty_like(steps[t]));
  }
  // steps as one packed fp32 tensor [T]
  auto steps_flat = at::empty({T}, params[0].options());
  {
    std::vector<at::Tensor> svec;
    for (auto& s : steps) svec.push_back(s.reshape({1}).to(at::kFloat));
    at::cat_out(steps_flat, svec, 0);
  }

  std::vector<AdamChunk> chunks;
  for (int t = 0; t < T; ++t) {
    long n = params[t].numel();
    for (long off = 0; off < n; off += CHUNK_ELEMS) {
      AdamChunk c;
      c.p = params[t].data_ptr<float>() + off;
      c.g = g_bf16 ? (const void*)((const bf16*)grads[t].data_ptr() + off)
                   : (const void*)(grads[t].data_ptr<float>() + off);
      c.ea = exp_avgs[t].data_ptr<float>() + off;
      c.eas = exp_avg_sqs[t].data_ptr<float>() + off;
      c.out_p = out_p[t].data_ptr<float>() + off;
      c.out_ea = out_ea[t].data_ptr<float>() + off;
      c.out_eas = out_eas[t].data_ptr<float>() + off;
      c.tensor_idx = t;
      c.n = (int)std::min<long>(CHUNK_ELEMS, n - off);
      chunks.push_back(c);
    }
  }
  // capture-safe table upload through the keyed staging pool (one pair
  // per parameter bank — the fuse pass emits a bf16-grad and an
  // fp32-grad fused step per optimizer).
  static std::map<std::pair<size_t, const void*>, StagingPair> g_pool;
  const size_t bytes = chunks.size() * sizeof(AdamChunk);
  const void* key = params[0].data_ptr();
  bool capturing = at::cuda::currentStreamCaptureStatusMayInitCtx() !=
                   at::cuda::CaptureStatus::None;
  auto& st = staging_for(g_pool, bytes, key, params[0].device(), capturing,
                         "fused_adam_step");
  memcpy(st.pinned.data_ptr(), chunks.data(), bytes);
  auto stream = at::cuda::getCurrentCUDAStream();
  C10_CUDA_CHECK(hipMemcpyAsync(st.dev.data_ptr(), st.pinned.data_ptr(),
                                bytes, hipMemcpyHostToDevice, stream));
  hipLaunchKernelGGL(fused_adam_kernel, dim3(chunks.size()), dim3(256), 0,
      stream, (const AdamChunk*)st.dev.data_ptr(), (int)chunks.size(),
      (int)g_bf16, steps_flat.data_ptr<float>(), (float)lr, (float)beta1,
      (float)beta2, (float)weight_decay, (float)eps);
  staging_finish(g_pool, bytes, key, capturing, stream);
  // bump steps on host-free path: out_step = step + 1
  for (int t = 0; t < T; ++t)
    out_steps[t] = steps[t] + 1;
  return {out_p, out_ea, out_eas, out_steps};
}

// ---------------------------------------------------------------- SGD -------
// Fused multi-tensor SGD(momentum) step: same chunk-table scheme as Adam
// (own persistent staging — the two kernels must not clobber each
// other's capture-recorded buffers).
struct SgdChunk {
  const float* p;
  const void* g;      // fp32 or bf16 (g_bf16 flag of the launch)
  const float* buf;   // nullptr when momentum == 0
  float* out_p;
  float* out_buf;     // nullptr when momentum == 0
  int n;
};

__global__ void fused_sgd_kernel(const SgdChunk* __restrict__ chunks,
                                 int n_chunks, int g_bf16, float lr,
                                 float momentum, float dampening,
                                 float weight_decay, int nesterov) {
  const int c = blockIdx.x;
  if (c >= n_chunks) return;
  const SgdChunk ch = chunks[c];
  for (int i = threadIdx.x; i < ch.n; i += blockDim.x) {
    float grad = g_bf16 ? bf2f(((const bf16*)ch.g)[i])
                        : ((const float*)ch.g)[i];
    if (weight_decay != 0.f) grad += weight_decay * ch.p[i];
    float upd = grad;
    if (momentum != 0.f) {
      float nb = momentum * ch.buf[i] + (1.f - dampening) * grad;
      ch.out_buf[i] = nb;
      upd = nesterov ? grad + momentum * nb : nb;
    }
    ch.out_p[i] = ch.p[i] - lr * upd;
  }
}

std::tuple<std::vector<at::Tensor>, std::vector<at::Tensor>>
fused_sgd_step(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
               std::vector<at::Tensor> bufs, double lr, double momentum,
               double dampening, double weight_decay, bool nesterov) {
  const int T = params.size();
  const bool has_m = momentum != 0.0;
  TORCH_CHECK(!has_m || (int)bufs.size() == T,
              "momentum needs one buffer per param");
  std::vector<at::Tensor> out_p, out_buf;
  out_p.reserve(T);
  const bool g_bf16 = !grads.empty()
      && grads[0].scalar_type() == at::kBFloat16;
  for (int t = 0; t < T; ++t) {
    TORCH_CHECK(params[t].scalar_type() == at::kFloat, "fp32 params only");
    TORCH_CHECK(grads[t].scalar_type()
                    == (g_bf16 ? at::kBFloat16 : at::kFloat),
                "grads must be uniformly fp32 or bf16");
    TORCH_CHECK(params[t].is_contiguous() && grads[t].is_contiguous());
    out_p.push_back(at::empty_like(params[t]));
    if (has_m) out_buf.push_back(at::empty_like(bufs[t]));
  }
  std::vector<SgdChunk> chunks;
  for (int t = 0; t < T; ++t) {
    long n = params[t].numel();
    for (long off = 0; off < n; off += CHUNK_ELEMS) {
      SgdChunk c;
      c.p = params[t].data_ptr<float>() + off;
      c.g = g_bf16 ? (const void*)((const bf16*)grads[t].data_ptr() + off)
                   : (const void*)(grads[t].data_ptr<float>() + off);
      c.buf = has_m ? bufs[t].data_ptr<float>() + off : nullptr;
      c.out_p = out_p[t].data_ptr<float>() + off;
      c.out_buf = has_m ? out_buf[t].data_ptr<float>() + off : nullptr;
      c.n = (int)std::min<long>(CHUNK_ELEMS, n - off);
      chunks.push_back(c);
    }
  }
  static std::map<std::pair<size_t, const void*>, StagingPair> g_pool;
  const size_t bytes = chunks.size() * sizeof(SgdChunk);
  const void* key = params[0].data_ptr();
  bool capturing = at::cuda::currentStreamCaptureStatusMayInitCtx() !=
                   at::cuda::CaptureStatus::None;
  auto& st = staging_for(g_pool, bytes, key, params[0].device(), capturing,
                         "fused_sgd_step");
  memcpy(st.pinned.data_ptr(), chunks.data(), bytes);
  auto stream = at::cuda::getCurrentCUDAStream();
  C10_CUDA_CHECK(hipMemcpyAsync(st.dev.data_ptr(), st.pinned.data_ptr(),
                                bytes, hipMemcpyHostToDevice, stream));
  hipLaunchKernelGGL(fused_sgd_kernel, dim3(chunks.size()), dim3(256), 0,
      stream, (const SgdChunk*)st.dev.data_ptr(), (int)chunks.size(),
      (int)g_bf16, (float)lr, (float)momentum, (float)dampening,
      (float)weight_decay, nesterov ? 1 : 0);
  staging_finish(g_pool, bytes, key, capturing, stream);
  return {out_p, out_buf};
}
