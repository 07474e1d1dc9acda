// Common helpers for room_amd CDNA4 (gfx950) kernels.
// Target: MI355X only — wave64, 256 CUs / 8 XCDs, LDS 160 KiB/CU, HBM3E 8 TB/s.
// No CUDA compatibility, no multi-arch dispatch.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define WAVE 64
#define DEV __device__ __forceinline__

// bf16 x8 = 16B vector load unit (coalescing sweet spot, guide G13)
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) short bf16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) float f32x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

DEV float bf2f(short u) {
  union { float f; uint32_t u32; } c;
  c.u32 = (uint32_t)(uint16_t)u << 16;
  return c.f;
}

DEV short f2bf(float f) {
  union { float f; uint32_t u32; } c;
  c.f = f;
  // round-to-nearest-even
  uint32_t lsb = (c.u32 >> 16) & 1;
  c.u32 += 0x7fff + lsb;
  return (short)(c.u32 >> 16);
}

// full-wave sum reduction (64 lanes)
DEV float wave_reduce_sum(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE);
  return v;
}

DEV float wave_reduce_max(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// block-level sum reduce across up to 16 waves; smem must be float[16]
DEV float block_reduce_sum(float v, float* smem) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  int nw = (blockDim.x + WAVE - 1) / WAVE;
  v = (lane < nw) ? smem[lane] : 0.0f;
  v = wave_reduce_sum(v);
  return __shfl(v, 0, WAVE);
}

DEV float block_reduce_max(float v, float* smem) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  int nw = (blockDim.x + WAVE - 1) / WAVE;
  v = (lane < nw) ? smem[lane] : -INFINITY;
  v = wave_reduce_max(v);
  return __shfl(v, 0, WAVE);
}

#define HIP_CHECK_KERNEL() do { \
    hipError_t e = hipGetLastError(); \
    if (e != hipSuccess) { \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e)); \
    } \
  } while (0)
