// Dense skinny-batch GEMV: y[B, N] = x[B, H] @ W[N, H]^T  (decode projections).
//
// hipBLASLt runs these M≤8 shapes at ~1.1 TB/s (measured, scripts/
// gpu_op_microbench.py); this kernel streams W once for ALL batch rows
// (x rows are L2-resident) with 16 B/lane coalesced loads. Used for the
// QKV/O/router/lm_head projections at decode; prefill keeps hipBLASLt.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

#define GEMV_MAXB 8

template <bool F32OUT>
__global__ __launch_bounds__(256)
void gemv_kernel(void* __restrict__ y,           // [B, N] bf16 or f32
                 const short* __restrict__ x,     // [B, H]
                 const short* __restrict__ w,     // [N, H]
                 int B, int H, int N) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = blockIdx.x * 4 + wid;
  if (n >= N) return;

  float acc[GEMV_MAXB];
  #pragma unroll
  for (int b = 0; b < GEMV_MAXB; ++b) acc[b] = 0.f;

  const short* wrow = w + (long)n * H;
  for (int base = lane * 8; base < H; base += WAVE * 8) {
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(wrow + base);
    float wf[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) wf[j] = bf2f(wv[j]);
    #pragma unroll
    for (int b = 0; b < GEMV_MAXB; ++b) {
      if (b < B) {
        bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + (long)b * H + base);
        #pragma unroll
        for (int j = 0; j < 8; ++j) acc[b] += wf[j] * bf2f(xv[j]);
      }
    }
  }
  #pragma unroll
  for (int b = 0; b < GEMV_MAXB; ++b) {
    if (b < B) {
      float r = wave_reduce_sum(acc[b]);
      if (lane == 0) {
        if (F32OUT)
          ((float*)y)[(long)b * N + n] = r;
        else
          ((short*)y)[(long)b * N + n] = f2bf(r);
      }
    }
  }
}

void gemv(torch::Tensor y, torch::Tensor x, torch::Tensor w) {
  const int B = x.size(0), H = x.size(1), N = w.size(0);
  TORCH_CHECK(B <= GEMV_MAXB, "gemv handles B<=8 (decode); use hipBLASLt above");
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && w.dtype() == torch::kBFloat16);
  TORCH_CHECK(H % (WAVE * 8) == 0, "H must be a multiple of 512");
  dim3 grid((N + 3) / 4), block(256);
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
