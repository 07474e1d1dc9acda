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

// BN = exact batch (compile-time): the generic MAXB=8 unroll carried 8 acc
// registers + per-lane bounds branches at any B — register pressure blocked
// software-pipelining of the 8 independent H-iterations.
template <bool F32OUT, int BN>
__global__ __launch_bounds__(256)
void gemv_kernel(void* __restrict__ y,           // [BN, N] bf16 or f32
                 const short* __restrict__ x,     // [BN, H]
                 const short* __restrict__ w,     // [N, H]
                 int H, int N) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = blockIdx.x * 4 + wid;
  if (n >= N) return;

  float acc[BN];
  #pragma unroll
  for (int b = 0; b < BN; ++b) acc[b] = 0.f;

  const short* wrow = w + (long)n * H;
  for (int base = lane * 8; base < H; base += WAVE * 8) {
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(wrow + base);
    float wf[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) wf[j] = bf2f(wv[j]);
    #pragma unroll
    for (int b = 0; b < BN; ++b) {
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + (long)b * H + base);
      #pragma unroll
      for (int j = 0; j < 8; ++j) acc[b] += wf[j] * bf2f(xv[j]);
    }
  }
  #pragma unroll
  for (int b = 0; b < BN; ++b) {
    float r = wave_reduce_sum(acc[b]);
    if (lane == 0) {
      if (F32OUT)
        ((float*)y)[(long)b * N + n] = r;
      else
        ((short*)y)[(long)b * N + n] = f2bf(r);
    }
  }
}

// x-rows staged through LDS once per WG: at B≥3 the per-wave x re-reads
// dominate L2 traffic (B=5, H=2048: each wave pulls 20 KB of x per dot →
// ~100 MB of L2 reads per QKV call across the 1280-WG grid; staging reads
// them once per WG). Dot loop is identical, sourced from LDS.
template <bool F32OUT, int BN>
__global__ __launch_bounds__(256)
void gemv_ldsx_kernel(void* __restrict__ y,           // [BN, N]
                      const short* __restrict__ x,     // [BN, H]
                      const short* __restrict__ w,     // [N, H]
                      int H, int N) {
  extern __shared__ short xs[];                        // [BN * H]
  for (int i = threadIdx.x * 8; i < BN * H; i += 256 * 8)
    *reinterpret_cast<bf16x8*>(&xs[i]) =
        *reinterpret_cast<const bf16x8*>(x + i);
  __syncthreads();

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = blockIdx.x * 4 + wid;
  if (n >= N) return;

  float acc[BN];
  #pragma unroll
  for (int b = 0; b < BN; ++b) acc[b] = 0.f;
  const short* wrow = w + (long)n * H;
  for (int base = lane * 8; base < H; base += WAVE * 8) {
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(wrow + base);
    float wf[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) wf[j] = bf2f(wv[j]);
    #pragma unroll
    for (int b = 0; b < BN; ++b) {
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(&xs[b * H + base]);
      #pragma unroll
      for (int j = 0; j < 8; ++j) acc[b] += wf[j] * bf2f(xv[j]);
    }
  }
  #pragma unroll
  for (int b = 0; b < BN; ++b) {
    float r = wave_reduce_sum(acc[b]);
    if (lane == 0) {
      if (F32OUT)
        ((float*)y)[(long)b * N + n] = r;
      else
        ((short*)y)[(long)b * N + n] = f2bf(r);
    }
  }
}

static bool gemv_ldsx_enabled() {
  static bool on = [] {
    const char* e = getenv("ROOMAMD_NO_GEMV_LDSX");
    return !(e && e[0] == '1');
  }();
  return on;
}

template <bool F32OUT>
static void gemv_launch(void* y, const short* x, const short* w, int B, int H,
                        int N, hipStream_t s) {
  dim3 grid((N + 3) / 4), block(256);
  const size_t lds = (size_t)B * H * sizeof(short);
  // LDS staging wins only on very wide-N shapes (lm_head: 152k rows,
  // 176 -> 151 us at B=5) where the grid's aggregate L2 x-traffic is the
  // bottleneck; on the 2-5k-row projections the barrier+occupancy cost
  // outweighs it (9.4 -> 9.9 us measured). Gate on N.
  const bool use_lds = gemv_ldsx_enabled() && B >= 3 && lds <= 64 * 1024
                       && N >= 16384;
  switch (B) {
#define GEMV_CASE(BN) \
    case BN: \
      if (use_lds) \
        hipLaunchKernelGGL((gemv_ldsx_kernel<F32OUT, BN>), grid, block, lds, \
                           s, y, x, w, H, N); \
      else \
        hipLaunchKernelGGL((gemv_kernel<F32OUT, BN>), grid, block, 0, s, \
                           y, x, w, H, N); \
      break;
    GEMV_CASE(1) GEMV_CASE(2) GEMV_CASE(3) GEMV_CASE(4)
    GEMV_CASE(5) GEMV_CASE(6) GEMV_CASE(7) GEMV_CASE(8)
#undef GEMV_CASE
    default: TORCH_CHECK(false, "gemv handles B<=8");
  }
}

void gemv(torch::Tensor y, torch::Tensor x, torch::Tensor w) {
  const int B = x.size(0), H = x.size(1), N = w.size(0);
  TORCH_CHECK(B >= 1 && B <= GEMV_MAXB,
              "gemv handles B<=8 (decode); use hipBLASLt above");
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && w.dtype() == torch::kBFloat16);
  TORCH_CHECK(H % (WAVE * 8) == 0, "H must be a multiple of 512");
  hipStream_t s = c10::hip::getCurrentHIPStream();
  if (y.dtype() == torch::kFloat32)
    gemv_launch<true>(y.data_ptr(), (const short*)x.data_ptr(),
                      (const short*)w.data_ptr(), B, H, N, s);
  else
    gemv_launch<false>(y.data_ptr(), (const short*)x.data_ptr(),
                       (const short*)w.data_ptr(), B, H, N, s);
  HIP_CHECK_KERNEL();
}

// ------------------------------------------------ fused add+RMSNorm+GEMV
// y[B,N] = rmsnorm(x + delta)·γ @ W^T, and (block 0 only) x_out = x + delta.
// Folds the per-layer fused_add_rmsnorm kernel and the normed-activation
// buffer round-trip into the projection itself: each wave already streams the
// full activation row, so the row norm is one extra L1-hot pass (decode chains
// ~14 dependent small kernels/layer; every removed kernel removes a
// launch+fill/drain bubble — see profiles/PERF_NOTES.md).
// Norm matches fused_add_rmsnorm semantics: computed over the bf16-rounded sum.
template <bool F32OUT, bool DELTA_F32, bool HAS_DELTA, int BN>
__global__ __launch_bounds__(256)
void gemv_addnorm_kernel(void* __restrict__ y,            // [BN, N]
                         const short* __restrict__ x,      // [BN, H]
                         const void* __restrict__ delta_,  // [BN, H] or null
                         short* __restrict__ x_out,        // [BN, H]
                         const short* __restrict__ gamma,  // [H]
                         const short* __restrict__ w,      // [N, H]
                         int H, int N, float eps) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = blockIdx.x * 4 + wid;

  // pass 1 (block-cooperative): row sums of squares, bf16-rounded to match
  // fused_add_rmsnorm. All 256 threads share the work (a per-wave pass was
  // measured as a B=5 regression: 4× redundant VALU across the block).
  __shared__ float ss_sh[BN];
  if (threadIdx.x < BN) ss_sh[threadIdx.x] = 0.f;
  __syncthreads();
  #pragma unroll
  for (int b = 0; b < BN; ++b) {
    {
      float ss = 0.f;
      for (int base = threadIdx.x * 8; base < H; base += 256 * 8) {
        bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + (long)b * H + base);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          float v = bf2f(xv[j]);
          if (HAS_DELTA) {
            if (DELTA_F32)
              v += ((const float*)delta_)[(long)b * H + base + j];
            else
              v += bf2f(((const short*)delta_)[(long)b * H + base + j]);
            v = bf2f(f2bf(v));
          }
          ss += v * v;
        }
      }
      ss = wave_reduce_sum(ss);
      if (lane == 0) atomicAdd(&ss_sh[b], ss);
    }
  }
  __syncthreads();
  float scale[BN];
  #pragma unroll
  for (int b = 0; b < BN; ++b)
    scale[b] = rsqrtf(ss_sh[b] / H + eps);

  // pass 1.5 (block-cooperative): stage xn = (x+δ)·scale·γ into LDS once, so
  // the dot loop is lean (w-load + LDS-load + FMA; no per-wave re-add/round)
  extern __shared__ short xn_sh[];  // [B][H] bf16
  #pragma unroll
  for (int b = 0; b < BN; ++b) {
    {
      for (int base = threadIdx.x * 8; base < H; base += 256 * 8) {
        bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + (long)b * H + base);
        bf16x8 gv = *reinterpret_cast<const bf16x8*>(gamma + base);
        bf16x8 o;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          float v = bf2f(xv[j]);
          if (HAS_DELTA) {
            if (DELTA_F32)
              v += ((const float*)delta_)[(long)b * H + base + j];
            else
              v += bf2f(((const short*)delta_)[(long)b * H + base + j]);
            v = bf2f(f2bf(v));
          }
          o[j] = f2bf(v * scale[b] * bf2f(gv[j]));
        }
        *reinterpret_cast<bf16x8*>(&xn_sh[b * H + base]) = o;
      }
    }
  }
  __syncthreads();

  // block 0 stores the updated residual (other blocks never read x_out here)
  if (HAS_DELTA && blockIdx.x == 0) {
    for (int i = threadIdx.x * 8; i < BN * H; i += 256 * 8) {
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + i);
      bf16x8 o;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf2f(xv[j]);
        if (DELTA_F32) v += ((const float*)delta_)[i + j];
        else v += bf2f(((const short*)delta_)[i + j]);
        o[j] = f2bf(v);
      }
      *reinterpret_cast<bf16x8*>(x_out + i) = o;
    }
  }
  if (n >= N) return;

  // pass 2: lean dot against the LDS-staged normed rows
  float acc[BN];
  #pragma unroll
  for (int b = 0; b < BN; ++b) acc[b] = 0.f;
  const short* wrow = w + (long)n * H;
  for (int base = lane * 8; base < H; base += WAVE * 8) {
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(wrow + base);
    float wf[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) wf[j] = bf2f(wv[j]);
    #pragma unroll
    for (int b = 0; b < BN; ++b) {
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(&xn_sh[b * H + base]);
      #pragma unroll
      for (int j = 0; j < 8; ++j) acc[b] += wf[j] * bf2f(xv[j]);
    }
  }
  #pragma unroll
  for (int b = 0; b < BN; ++b) {
    float r = wave_reduce_sum(acc[b]);
    if (lane == 0) {
      if (F32OUT) ((float*)y)[(long)b * N + n] = r;
      else ((short*)y)[(long)b * N + n] = f2bf(r);
    }
  }
}

void gemv_addnorm(torch::Tensor y, torch::Tensor x, torch::Tensor delta,
                  torch::Tensor x_out, torch::Tensor gamma, torch::Tensor w,
                  double eps) {
  const int B = x.size(0), H = x.size(1), N = w.size(0);
  TORCH_CHECK(B <= GEMV_MAXB && H % (WAVE * 8) == 0);
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && w.dtype() == torch::kBFloat16);
  const bool has_delta = delta.numel() > 0;
  const bool delta_f32 = has_delta && delta.dtype() == torch::kFloat32;
  const bool f32out = y.dtype() == torch::kFloat32;
  dim3 grid((N + 3) / 4), block(256);
  const size_t lds = (size_t)B * H * sizeof(short);  // staged normed rows
  TORCH_CHECK(lds <= 160 * 1024, "B*H too large for LDS staging");
  hipStream_t s = c10::hip::getCurrentHIPStream();
  const void* dptr = has_delta ? delta.data_ptr() : nullptr;
  #define LAUNCH_AN1(FO, DF, HD, BN) hipLaunchKernelGGL( \
      (gemv_addnorm_kernel<FO, DF, HD, BN>), grid, block, lds, s, y.data_ptr(), \
      (const short*)x.data_ptr(), dptr, (short*)x_out.data_ptr(), \
      (const short*)gamma.data_ptr(), (const short*)w.data_ptr(), \
      H, N, (float)eps)
  #define LAUNCH_AN(FO, DF, HD) do { switch (B) { \
    case 1: LAUNCH_AN1(FO, DF, HD, 1); break; \
    case 2: LAUNCH_AN1(FO, DF, HD, 2); break; \
    case 3: LAUNCH_AN1(FO, DF, HD, 3); break; \
    case 4: LAUNCH_AN1(FO, DF, HD, 4); break; \
    case 5: LAUNCH_AN1(FO, DF, HD, 5); break; \
    case 6: LAUNCH_AN1(FO, DF, HD, 6); break; \
    case 7: LAUNCH_AN1(FO, DF, HD, 7); break; \
    default: LAUNCH_AN1(FO, DF, HD, 8); break; } } while (0)
  if (!has_delta) {
    if (f32out) LAUNCH_AN(true, false, false); else LAUNCH_AN(false, false, false);
  } else if (delta_f32) {
    if (f32out) LAUNCH_AN(true, true, true); else LAUNCH_AN(false, true, true);
  } else {
    if (f32out) LAUNCH_AN(true, false, true); else LAUNCH_AN(false, false, true);
  }
  #undef LAUNCH_AN
  #undef LAUNCH_AN1
  HIP_CHECK_KERNEL();
}
