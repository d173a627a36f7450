// Shared helpers for the gfx950 kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

using bf16 = __hip_bfloat16;

// 8 bf16 = 16 bytes: the coalescing sweet spot on CDNA4
// (cdna_hip_programming.md Guideline 13)
struct alignas(16) bf16x8 { bf16 v[8]; };
struct alignas(16) f32x4v { float x, y, z, w; };

DEVINL float bf2f(bf16 x) { return __bfloat162float(x); }
DEVINL bf16 f2bf(float x) { return __float2bfloat16(x); }

// wave-wide reductions over 64 lanes
DEVINL float wave_sum(float x) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_down(x, off, WAVE);
  return __shfl(x, 0, WAVE);
}

DEVINL float wave_max(float x) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_down(x, off, WAVE));
  return __shfl(x, 0, WAVE);
}


// wait for LDS (lgkm) traffic only — unlike s_waitcnt(0) this does NOT
// flush vmcnt, so in-flight global prefetch loads keep overlapping.
// imm encoding (gfx90a+): vmcnt=63 (bits 3:0 + 15:14), expcnt=7 (6:4),
// lgkmcnt=0 (13:8)
DEVINL void lds_fence() { __builtin_amdgcn_s_waitcnt(0xC07F); }
