// Dense skinny-batch GEMV: y[B, N] = x[B, H] @ W[N, H]^T  (decode projections).
//
// hipBLASLt runs these M≤8 shapes at ~1.1 TB/s (measured, scripts/
// gpu_op_microbench.py). v1 here (wave-per-output) reached ~2.2 TB/s but its
// waves lived only 4 iterations — launch/drain-bound. v2: each wave owns FOUR
// consecutive output rows, streaming their W rows together; x rows (L2-hot)
// are unpacked once per k-chunk and reused across the 4 rows.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

#define GEMV_MAXB 8
#define GEMV_ROWS 4   // output rows per wave

template <bool F32OUT>
__global__ __launch_bounds__(256)
void gemv_kernel(void* __restrict__ y,           // [B, N] bf16 or f32
                 const short* __restrict__ x,     // [B, H]
                 const short* __restrict__ w,     // [N, H]
                 int B, int H, int N) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n0 = (blockIdx.x * 4 + wid) * GEMV_ROWS;
  if (n0 >= N) return;

  float acc[GEMV_ROWS][GEMV_MAXB];
  #pragma unroll
  for (int r = 0; r < GEMV_ROWS; ++r)
    #pragma unroll
    for (int b = 0; b < GEMV_MAXB; ++b) acc[r][b] = 0.f;

  for (int base = lane * 8; base < H; base += WAVE * 8) {
    float xf[GEMV_MAXB][8];
    #pragma unroll
    for (int b = 0; b < GEMV_MAXB; ++b) {
      if (b < B) {
        bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + (long)b * H + base);
        #pragma unroll
        for (int j = 0; j < 8; ++j) xf[b][j] = bf2f(xv[j]);
      }
    }
    #pragma unroll
    for (int r = 0; r < GEMV_ROWS; ++r) {
      const int n = n0 + r;
      if (n >= N) break;
      bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + (long)n * H + base);
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float wf = bf2f(wv[j]);
        #pragma unroll
        for (int b = 0; b < GEMV_MAXB; ++b)
          if (b < B) acc[r][b] += wf * xf[b][j];
      }
    }
  }
  #pragma unroll
  for (int r = 0; r < GEMV_ROWS; ++r) {
    const int n = n0 + r;
    if (n >= N) break;
    #pragma unroll
    for (int b = 0; b < GEMV_MAXB; ++b) {
      if (b < B) {
        float v = wave_reduce_sum(acc[r][b]);
        if (lane == 0) {
          if (F32OUT)
            ((float*)y)[(long)b * N + n] = v;
          else
            ((short*)y)[(long)b * N + n] = f2bf(v);
        }
      }
    }
  }
}

void gemv(torch::Tensor y, torch::Tensor x, torch::Tensor w) {
  const int B = x.size(0), H = x.size(1), N = w.size(0);
  TORCH_CHECK(B <= GEMV_MAXB, "gemv handles B<=8 (decode); use hipBLASLt above");
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && w.dtype() == torch::kBFloat16);
  TORCH_CHECK(H % (WAVE * 8) == 0, "H must be a multiple of 512");
  dim3 grid((N + 4 * GEMV_ROWS - 1) / (4 * GEMV_ROWS)), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  if (y.dtype() == torch::kFloat32) {
    hipLaunchKernelGGL(gemv_kernel<true>, grid, block, 0, s,
                       y.data_ptr(), (const short*)x.data_ptr(),
                       (const short*)w.data_ptr(), B, H, N);
  } else {
    hipLaunchKernelGGL(gemv_kernel<false>, grid, block, 0, s,
                       y.data_ptr(), (const short*)x.data_ptr(),
                       (const short*)w.data_ptr(), B, H, N);
  }
  HIP_CHECK_KERNEL();
}
