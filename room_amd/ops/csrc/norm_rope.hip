// RMSNorm (+fused residual-add) and fused per-head QK-RMSNorm + RoPE.
//
// The reference delegated all model math to Ollama (src/shared/local-model.ts);
// these are the in-process CDNA4 replacements. All memory-bound: bf16 loads
// vectorized 8-wide (16 B/lane), fp32 accumulation, one workgroup per row.
// cos/sin tables are host-precomputed (on-device trig turns memory-bound into
// VALU-bound — guide Appendix B).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

// ---------------------------------------------------------------- rmsnorm
// y = x / rms(x) * w.   rows × cols, cols % 8 == 0, cols <= 8192.

__global__ void rmsnorm_kernel(short* __restrict__ out,
                               const short* __restrict__ in,
                               const short* __restrict__ weight,
                               int cols, float eps) {
  __shared__ float red[16];
  const long row = blockIdx.x;
  const short* x = in + row * (long)cols;
  short* y = out + row * (long)cols;

  float sumsq = 0.f;
  const int vecs = cols / 8;
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + i * 8);
    #pragma unroll
    for (int j = 0; j < 8; ++j) { float f = bf2f(v[j]); sumsq += f * f; }
  }
  float total = block_reduce_sum(sumsq, red);
  float scale = rsqrtf(total / cols + eps);

  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + i * 8);
    bf16x8 w = *reinterpret_cast<const bf16x8*>(weight + i * 8);
    bf16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(v[j]) * scale * bf2f(w[j]));
    *reinterpret_cast<bf16x8*>(y + i * 8) = o;
  }
}

// residual = residual + x;  y = rmsnorm(residual) * w   (fused: one HBM pass)
// IN_F32: accepts the MoE f32 accumulator directly (skips a cast kernel)
template <bool IN_F32>
__global__ void fused_add_rmsnorm_kernel(short* __restrict__ out,
                                         short* __restrict__ residual,
                                         const void* __restrict__ in_,
                                         const short* __restrict__ weight,
                                         int cols, float eps) {
  __shared__ float red[16];
  const long row = blockIdx.x;
  const short* xb = IN_F32 ? nullptr : (const short*)in_ + row * (long)cols;
  const float* xf = IN_F32 ? (const float*)in_ + row * (long)cols : nullptr;
  short* r = residual + row * (long)cols;
  short* y = out + row * (long)cols;

  float sumsq = 0.f;
  const int vecs = cols / 8;
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    bf16x8 rv = *reinterpret_cast<const bf16x8*>(r + i * 8);
    bf16x8 nr;
    float xv[8];
    if (IN_F32) {
      f32x4 a = *reinterpret_cast<const f32x4*>(xf + i * 8);
      f32x4 b = *reinterpret_cast<const f32x4*>(xf + i * 8 + 4);
      #pragma unroll
      for (int j = 0; j < 4; ++j) { xv[j] = a[j]; xv[4 + j] = b[j]; }
    } else {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(xb + i * 8);
      #pragma unroll
      for (int j = 0; j < 8; ++j) xv[j] = bf2f(v[j]);
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = xv[j] + bf2f(rv[j]);
      nr[j] = f2bf(f);
      float g = bf2f(nr[j]);         // norm over the *stored* bf16 residual
      sumsq += g * g;
    }
    *reinterpret_cast<bf16x8*>(r + i * 8) = nr;
  }
  float total = block_reduce_sum(sumsq, red);
  float scale = rsqrtf(total / cols + eps);

  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    bf16x8 rv = *reinterpret_cast<const bf16x8*>(r + i * 8);
    bf16x8 w = *reinterpret_cast<const bf16x8*>(weight + i * 8);
    bf16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(rv[j]) * scale * bf2f(w[j]));
    *reinterpret_cast<bf16x8*>(y + i * 8) = o;
  }
}

// ------------------------------------------------ fused QK-norm + RoPE
// Qwen3 applies per-head RMSNorm to Q and K, then rotary embedding.
// One fused kernel: grid = (tokens, q_heads + kv_heads); each workgroup is
// TWO waves (head_dim=128 → one f32/lane per half). rotate-half (NeoX) form:
//   out[i]      = x[i]   * cos[i] - x[i+D/2] * sin[i]
//   out[i+D/2]  = x[i+D/2] * cos[i] + x[i]   * sin[i]
// cos/sin: [max_pos, D/2] fp32, positions: [tokens] int32.

__global__ void qk_norm_rope_kernel(short* __restrict__ q,      // [T, Hq*D]
                                    short* __restrict__ k,      // [T, Hk*D]
                                    const short* __restrict__ q_w,  // [D]
                                    const short* __restrict__ k_w,  // [D]
                                    const float* __restrict__ cos_t,  // [P, D/2]
                                    const float* __restrict__ sin_t,
                                    const int* __restrict__ pos,      // [T]
                                    int n_qheads, int n_kvheads,
                                    int head_dim, float eps) {
  const int t = blockIdx.x;
  const int h = blockIdx.y;
  const bool is_q = h < n_qheads;
  short* base = is_q ? (q + ((long)t * n_qheads + h) * head_dim)
                     : (k + ((long)t * n_kvheads + (h - n_qheads)) * head_dim);
  const short* w = is_q ? q_w : k_w;

  // 128 threads, head_dim = 128: lane i owns element i.
  const int i = threadIdx.x;
  float x = bf2f(base[i]);
  // RMS over the head: two-wave reduce via LDS
  __shared__ float red[16];
  float total = block_reduce_sum(x * x, red);
  float scale = rsqrtf(total / head_dim + eps);
  float xn = x * scale * bf2f(w[i]);

  // exchange halves for rotate-half through LDS
  __shared__ float sh[128];
  sh[i] = xn;
  __syncthreads();
  const int half = head_dim / 2;
  const int p = pos[t];
  float out;
  if (i < half) {
    float c = cos_t[(long)p * half + i];
    float s = sin_t[(long)p * half + i];
    out = xn * c - sh[i + half] * s;
  } else {
    float c = cos_t[(long)p * half + (i - half)];
    float s = sin_t[(long)p * half + (i - half)];
    out = xn * c + sh[i - half] * s;
  }
  base[i] = f2bf(out);
}

// ---------------------------------------------------------------- silu_mul
// h = silu(gate) * up, operating on packed [rows, 2*inter] where first half
// is gate, second is up (MoE activation). Memory-bound, 8-wide.
__global__ void silu_mul_kernel(short* __restrict__ out,        // [rows, inter]
                                const short* __restrict__ gateup,  // [rows, 2*inter]
                                int inter) {
  const long row = blockIdx.x;
  const short* g = gateup + row * (long)(2 * inter);
  const short* u = g + inter;
  short* y = out + row * (long)inter;
  const int vecs = inter / 8;
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(g + i * 8);
    bf16x8 uv = *reinterpret_cast<const bf16x8*>(u + i * 8);
    bf16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(gv[j]);
      float s = gf / (1.0f + __expf(-gf));
      o[j] = f2bf(s * bf2f(uv[j]));
    }
    *reinterpret_cast<bf16x8*>(y + i * 8) = o;
  }
}

// ---------------------------------------------------------------- host wrappers

void rmsnorm(torch::Tensor out, torch::Tensor in, torch::Tensor weight, double eps) {
  TORCH_CHECK(in.is_cuda() && in.dtype() == torch::kBFloat16);
  TORCH_CHECK(in.is_contiguous() && out.is_contiguous());
  long rows = in.numel() / in.size(-1);
  int cols = in.size(-1);
  TORCH_CHECK(cols % 8 == 0);
  dim3 grid(rows), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsnorm_kernel, grid, block, 0, s,
                     (short*)out.data_ptr(), (const short*)in.data_ptr(),
                     (const short*)weight.data_ptr(), cols, (float)eps);
  HIP_CHECK_KERNEL();
}

void fused_add_rmsnorm(torch::Tensor out, torch::Tensor residual, torch::Tensor in,
                       torch::Tensor weight, double eps) {
  TORCH_CHECK(in.is_cuda());
  long rows = in.numel() / in.size(-1);
  int cols = in.size(-1);
  TORCH_CHECK(cols % 8 == 0);
  dim3 grid(rows), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  if (in.dtype() == torch::kFloat32) {
    hipLaunchKernelGGL(fused_add_rmsnorm_kernel<true>, grid, block, 0, s,
                       (short*)out.data_ptr(), (short*)residual.data_ptr(),
                       in.data_ptr(), (const short*)weight.data_ptr(),
                       cols, (float)eps);
  } else {
    TORCH_CHECK(in.dtype() == torch::kBFloat16);
    hipLaunchKernelGGL(fused_add_rmsnorm_kernel<false>, grid, block, 0, s,
                       (short*)out.data_ptr(), (short*)residual.data_ptr(),
                       in.data_ptr(), (const short*)weight.data_ptr(),
                       cols, (float)eps);
  }
  HIP_CHECK_KERNEL();
}

void qk_norm_rope(torch::Tensor q, torch::Tensor k, torch::Tensor q_w,
                  torch::Tensor k_w, torch::Tensor cos_t, torch::Tensor sin_t,
                  torch::Tensor positions, int64_t n_qheads, int64_t n_kvheads,
                  int64_t head_dim, double eps) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(head_dim == 128, "kernel assumes head_dim=128");
  int T = positions.size(0);
  dim3 grid(T, n_qheads + n_kvheads), block(head_dim);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(qk_norm_rope_kernel, grid, block, 0, s,
                     (short*)q.data_ptr(), (short*)k.data_ptr(),
                     (const short*)q_w.data_ptr(), (const short*)k_w.data_ptr(),
                     cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                     positions.data_ptr<int>(), (int)n_qheads, (int)n_kvheads,
                     (int)head_dim, (float)eps);
  HIP_CHECK_KERNEL();
}

void silu_mul(torch::Tensor out, torch::Tensor gateup) {
  TORCH_CHECK(gateup.is_cuda() && gateup.dtype() == torch::kBFloat16);
  long rows = gateup.numel() / gateup.size(-1);
  int inter = gateup.size(-1) / 2;
  TORCH_CHECK(inter % 8 == 0);
  dim3 grid(rows), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(silu_mul_kernel, grid, block, 0, s,
                     (short*)out.data_ptr(), (const short*)gateup.data_ptr(), inter);
  HIP_CHECK_KERNEL();
}

// ------------------------------------ fused QK-norm + RoPE + KV-cache write
// One launch replaces qk_norm_rope + write_kv: Q is normed+rotated in place;
// K is normed+rotated and scattered straight into the paged cache; V is
// scattered without a separate pass. grid (T, Hq + 2*Hk), 128 threads.
// qkv is the packed projection output [T, (Hq+2Hk)*D]; row_stride lets the
// kernel read q/k/v in place (no .contiguous() copies — profiled at 0.67
// ms/decode-step). Rotated Q is written back into the qkv buffer.
__global__ void qk_rope_write_kv_kernel(short* __restrict__ qkv,
                                        short* __restrict__ kcache,  // [NB,Hk,16,D]
                                        short* __restrict__ vcache,
                                        const short* __restrict__ q_w,
                                        const short* __restrict__ k_w,
                                        const float* __restrict__ cos_t,
                                        const float* __restrict__ sin_t,
                                        const int* __restrict__ block_table,
                                        const int* __restrict__ seq_ids,
                                        const int* __restrict__ pos,
                                        int n_qheads, int n_kvheads,
                                        int head_dim, int max_blocks,
                                        int row_stride, float eps) {
  const int t = blockIdx.x;
  const int h = blockIdx.y;
  const int i = threadIdx.x;
  const int p = pos[t];
  const int blk = block_table[(long)seq_ids[t] * max_blocks + p / 16];
  const long cache_off = ((long)blk * n_kvheads) * 16 * head_dim
                         + (long)(p % 16) * head_dim;

  const long row = (long)t * row_stride;
  if (h >= n_qheads + n_kvheads) {        // V scatter (no rope)
    const int hv = h - n_qheads - n_kvheads;
    vcache[cache_off + (long)hv * 16 * head_dim + i] =
        qkv[row + (long)(n_qheads + n_kvheads + hv) * head_dim + i];
    return;
  }

  const bool is_q = h < n_qheads;
  const short* src = qkv + row + (long)h * head_dim;  // q and k are packed
  const short* w = is_q ? q_w : k_w;

  float x = bf2f(src[i]);
  __shared__ float red[16];
  float total = block_reduce_sum(x * x, red);
  float scale = rsqrtf(total / head_dim + eps);
  float xn = x * scale * bf2f(w[i]);

  __shared__ float sh[128];
  sh[i] = xn;
  __syncthreads();
  const int half = head_dim / 2;
  float out;
  if (i < half) {
    out = xn * cos_t[(long)p * half + i] - sh[i + half] * sin_t[(long)p * half + i];
  } else {
    out = xn * cos_t[(long)p * half + (i - half)]
          + sh[i - half] * sin_t[(long)p * half + (i - half)];
  }
  if (is_q) {
    qkv[row + (long)h * head_dim + i] = f2bf(out);
  } else {
    const int hk = h - n_qheads;
    kcache[cache_off + (long)hk * 16 * head_dim + i] = f2bf(out);
  }
}

void qk_rope_write_kv(torch::Tensor qkv, torch::Tensor kcache,
                      torch::Tensor vcache, torch::Tensor q_w, torch::Tensor k_w,
                      torch::Tensor cos_t, torch::Tensor sin_t,
                      torch::Tensor block_table, torch::Tensor seq_ids,
                      torch::Tensor positions, int64_t n_qheads, double eps) {
  const int T = positions.size(0);
  const int n_kvheads = kcache.size(1);
  const int head_dim = kcache.size(3);
  TORCH_CHECK(head_dim == 128 && kcache.size(2) == 16);
  TORCH_CHECK(qkv.dim() == 2 && qkv.is_contiguous());
  const int row_stride = qkv.size(1);
  TORCH_CHECK(row_stride == (n_qheads + 2 * n_kvheads) * head_dim);
  const int max_blocks = block_table.size(1);
  dim3 grid(T, n_qheads + 2 * n_kvheads), block(head_dim);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(qk_rope_write_kv_kernel, grid, block, 0, s,
                     (short*)qkv.data_ptr(), (short*)kcache.data_ptr(),
                     (short*)vcache.data_ptr(), (const short*)q_w.data_ptr(),
                     (const short*)k_w.data_ptr(), cos_t.data_ptr<float>(),
                     sin_t.data_ptr<float>(), block_table.data_ptr<int>(),
                     seq_ids.data_ptr<int>(), positions.data_ptr<int>(),
                     (int)n_qheads, n_kvheads, head_dim, max_blocks,
                     row_stride, (float)eps);
  HIP_CHECK_KERNEL();
}
