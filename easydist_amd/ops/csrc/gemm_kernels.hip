// Hand-written MFMA bf16 GEMM for gfx950: C[M,N] = A[M,K] @ B[N,K]^T.
//
// The NT layout (both operands K-contiguous) is what traced linear layers
// produce: addmm(bias, x, t(W)) — the lowering pass rewrites mm(x, t(W))
// to this kernel, leaving NN/TN shapes on hipBLASLt.
//
// Structure (cdna_hip_programming.md §5, ladder step 3): 128x128 tile,
// 4 waves as a 2x2 wave grid (64x64 per wave), 4x4 accumulator fragments
// of v_mfma_f32_16x16x32_bf16, K-step 32, double-buffered LDS filled by
// 16-byte global_load_lds, XCD-aware bijective block swizzle (T1).
#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;

#define BM 128
#define BN 128
#define BK 32
#define N_XCD 8

// global_load_lds: LDS dest is wave-uniform base + lane*16 (guide §5 note)
DEVINL void glds16(const void* gsrc, void* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds_dst, 16, 0, 0);
}

extern "C" __global__ void __launch_bounds__(256)
gemm_nt_bf16(const bf16* __restrict__ A, const bf16* __restrict__ Bt,
             bf16* __restrict__ C, const bf16* __restrict__ bias,
             int M, int N, int K) {
  // dynamic LDS: [2 buffers][A 128x32 | B 128x32] bf16
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const unsigned ATILE = BM * BK * 2;          // bytes per A tile (8 KiB)
  const unsigned BUF = 2 * ATILE;              // A+B per buffer (16 KiB)

  // XCD-aware bijective swizzle of the workgroup id (T1)
  unsigned nwg_m = (M + BM - 1) / BM, nwg_n = (N + BN - 1) / BN;
  unsigned nwg = nwg_m * nwg_n;
  unsigned wg = blockIdx.x;
  {
    unsigned q = nwg / N_XCD, r = nwg % N_XCD;
    unsigned xcd = wg % N_XCD, idx = wg / N_XCD;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const unsigned m0 = (wg / nwg_n) * BM;
  const unsigned n0 = (wg % nwg_n) * BN;

  const int t = threadIdx.x;
  const int lane = t % WAVE;
  const int wave = t / WAVE;
  const int wm = (wave / 2) * 64;   // wave row offset in tile
  const int wn = (wave % 2) * 64;   // wave col offset

  // stage functions: 256 threads x 16 B = 4 KiB per pass; 2 passes per tile
  auto stage = [&](int buf, int k0) {
    // A tile: rows m0..m0+127, cols k0..k0+31, row-major [128][32]
    #pragma unroll
    for (int p = 0; p < 2; ++p) {
      int e = (t + p * 256) * 8;              // element offset in tile
      int row = e / BK, col = e % BK;
      const bf16* src = A + (long)(m0 + row) * K + k0 + col;
      // lds base for this WAVE's 1 KiB slice (dest = base + lane*16)
      glds16(src, smem + buf * BUF + (p * 256 + wave * 64) * 16);
    }
    #pragma unroll
    for (int p = 0; p < 2; ++p) {
      int e = (t + p * 256) * 8;
      int row = e / BK, col = e % BK;
      const bf16* src = Bt + (long)(n0 + row) * K + k0 + col;
      glds16(src, smem + buf * BUF + ATILE + (p * 256 + wave * 64) * 16);
    }
  };

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j)
      acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  stage(0, 0);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  const int frag_row = lane & 15;        // row inside 16-row fragment
  const int frag_k = (lane >> 4) * 8;    // k offset (8 contiguous bf16)

  int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    int buf = kt & 1;
    if (kt + 1 < nk) stage(buf ^ 1, (kt + 1) * BK);
    // fragment reads: A row (wm + i*16 + frag_row), k = frag_k..+8
    bf16x8v a_frag[4], b_frag[4];
    const char* base = smem;   // LDS base
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      unsigned off = buf * BUF + ((wm + i * 16 + frag_row) * BK + frag_k) * 2;
      a_frag[i] = *reinterpret_cast<const bf16x8v*>(base + off);
    }
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      unsigned off = buf * BUF + ATILE +
                     ((wn + j * 16 + frag_row) * BK + frag_k) * 2;
      b_frag[j] = *reinterpret_cast<const bf16x8v*>(base + off);
    }
    #pragma unroll
    for (int i = 0; i < 4; ++i)
      #pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
  }

  // epilogue: C/D layout col = lane&15, row = (lane>>4)*4 + reg
  const int c_col = lane & 15;
  const int c_row0 = (lane >> 4) * 4;
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      #pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int gm = m0 + wm + i * 16 + c_row0 + reg;
        int gn = n0 + wn + j * 16 + c_col;
        if (gm < M && gn < N) {
          float v = acc[i][j][reg];
          if (bias) v += bf2f(bias[gn]);
          C[(long)gm * N + gn] = f2bf(v);
        }
      }
    }
  }
}

at::Tensor gemm_nt(const at::Tensor& a, const at::Tensor& bt,
                   const std::optional<at::Tensor>& bias) {
  TORCH_CHECK(a.dtype() == at::kBFloat16 && bt.dtype() == at::kBFloat16);
  TORCH_CHECK(a.is_contiguous() && bt.is_contiguous(),
              "gemm_nt wants K-contiguous operands");
  const int M = a.size(0), K = a.size(1), N = bt.size(0);
  TORCH_CHECK(bt.size(1) == K);
  TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK == 0,
              "gemm_nt v1: M,N multiples of 128, K multiple of 32");
  auto c = at::empty({M, N}, a.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  unsigned nwg = (M / BM) * (N / BN);
  size_t lds = 2 * 2 * BM * BK * 2;   // 32 KiB
  hipLaunchKernelGGL(gemm_nt_bf16, dim3(nwg), dim3(256), lds, stream,
      (const bf16*)a.data_ptr(), (const bf16*)bt.data_ptr(),
      (bf16*)c.data_ptr(),
      bias ? (const bf16*)bias->data_ptr() : nullptr, M, N, K);
  return c;
}
