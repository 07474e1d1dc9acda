// Decode-path fusion kernels (round-2: collapse the ~11 dependent launches
// per layer; each removed kernel removes a launch + grid fill/drain bubble —
// profiles/PERF_NOTES.md puts the per-layer bubble cost at 30-40 µs).
//
// 1) attn_merge_o: split-KV partial merge + O-projection in ONE kernel.
//    The standalone merge (10 µs) writes a bf16 [T,Hq,D] buffer that the
//    O-GEMV (9.4 µs) immediately re-reads. Here each workgroup owns one
//    (kv-merge of one q-head) × (one 128-wide output tile of Wo): it merges
//    the head's NSPLITS partials into LDS (f32, no bf16 hop), stages its Wo
//    tile through LDS (each Wo byte is read exactly once across the grid),
//    and atomically accumulates the per-head contribution into the f32
//    output accumulator — the same f32-delta the fused add-norm consumes.
//
// 2) router_addnorm: residual-add + RMSNorm + router logits in ONE kernel,
//    also emitting the updated residual and the normed activations (the MoE
//    GEMV input). Replaces fused_add_rmsnorm + H-split router partial GEMV.
//    N=128 outputs → 32 workgroups; each does the (cheap, L2-hot) norm pass
//    into LDS and four 2048-wide dots per wave.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

#define HEAD_DIM 128
#define NSPLITS_MAX 64      // partial-buffer stride (paged_attn.hip)
#define AMO_NT 128          // output-tile width of attn_merge_o

// ------------------------------------------------------------ attn_merge_o
// grid: (N / AMO_NT, Hq); block 256.
// o_accum[b, n] += sum_d Wo[n, hq*128+d] * merged[b, hq, d]
template <int BN>
__global__ __launch_bounds__(256)
void attn_merge_o_kernel(float* __restrict__ o_accum,      // [BN, N] (pre-zeroed)
                         const float* __restrict__ part,    // [BN, Hq, NS, D]
                         const float* __restrict__ part_ml, // [BN, Hq, NS, 2]
                         const short* __restrict__ wo,      // [N, Hq*D]
                         int n_qheads, int N, int n_splits) {
  const int n0 = blockIdx.x * AMO_NT;
  const int hq = blockIdx.y;
  const int tid = threadIdx.x;

  __shared__ float merged[BN][HEAD_DIM + 4];
  // +2 bf16 row pad → 260 B row stride = 65 banks: consecutive rows land on
  // consecutive banks, so the 32 lanes reading distinct rows never conflict
  __shared__ short w_s[AMO_NT][HEAD_DIM + 2];

  // ---- phase A: merge this head's NSPLITS partials for every batch row.
  // All 256 threads cooperate per row: thread (sg = tid>>5, d4 = (tid&31)*4)
  // covers split group sg (NSPLITS/8 = 4 splits) × one f32x4 of dims — the
  // 4 loads per thread are independent (one latency window, vs a 32-long
  // strided chain in the first version); the 8 split-group partials reduce
  // through LDS.
  __shared__ float red[8][32][4];

  // issue the Wo tile loads FIRST (registers), so the HBM latency overlaps
  // phase A's merge work; the LDS store + barrier happen after phase A
  bf16x8 wreg[8];
  {
    const int r = tid >> 1;
    const int half = (tid & 1) * 64;              // 64 bf16 = 128 B
    const short* wrow = wo + ((long)(n0 + r) * n_qheads + hq) * HEAD_DIM + half;
    #pragma unroll
    for (int v8 = 0; v8 < 8; ++v8)
      wreg[v8] = *reinterpret_cast<const bf16x8*>(wrow + v8 * 8);
  }

  {
    const int sg = tid >> 5;                       // split group 0..7
    const int d32 = tid & 31;
    const int d4 = d32 * 4;
    const int SPG = n_splits / 8;                  // splits per group
    #pragma unroll
    for (int b = 0; b < BN; ++b) {
      const float* ml = part_ml + (((long)b * n_qheads + hq) * NSPLITS_MAX) * 2;
      const float* pacc = part + (((long)b * n_qheads + hq) * NSPLITS_MAX) * HEAD_DIM;
      float m_star = -INFINITY;
      #pragma unroll 8
      for (int s = 0; s < n_splits; ++s) m_star = fmaxf(m_star, ml[2 * s]);
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      for (int i = 0; i < SPG; ++i) {
        const int s = sg * SPG + i;
        const float ms = ml[2 * s];
        if (ms == -INFINITY) continue;
        const float f = __expf(ms - m_star);
        const f32x4 v = *reinterpret_cast<const f32x4*>(
            pacc + (long)s * HEAD_DIM + d4);
        acc[0] += v[0] * f; acc[1] += v[1] * f;
        acc[2] += v[2] * f; acc[3] += v[3] * f;
      }
      red[sg][d32][0] = acc[0]; red[sg][d32][1] = acc[1];
      red[sg][d32][2] = acc[2]; red[sg][d32][3] = acc[3];
      __syncthreads();
      // l_tot: every thread recomputes the full sum from ml (L1-hot, cheap)
      if (sg == 0) {
        float lt = 0.f;
        #pragma unroll 8
        for (int s = 0; s < n_splits; ++s) {
          const float ms = ml[2 * s];
          if (ms != -INFINITY) lt += ml[2 * s + 1] * __expf(ms - m_star);
        }
        const float inv = (lt > 0.f) ? 1.0f / lt : 0.f;
        f32x4 tot = {0.f, 0.f, 0.f, 0.f};
        #pragma unroll
        for (int g = 0; g < 8; ++g) {
          tot[0] += red[g][d32][0]; tot[1] += red[g][d32][1];
          tot[2] += red[g][d32][2]; tot[3] += red[g][d32][3];
        }
        merged[b][d4 + 0] = tot[0] * inv;
        merged[b][d4 + 1] = tot[1] * inv;
        merged[b][d4 + 2] = tot[2] * inv;
        merged[b][d4 + 3] = tot[3] * inv;
      }
      __syncthreads();
    }
  }

  // ---- stage the Wo tile from registers (loads issued before phase A)
  {
    const int r = tid >> 1;
    const int half = (tid & 1) * 64;
    #pragma unroll
    for (int v8 = 0; v8 < 8; ++v8)
      *reinterpret_cast<bf16x8*>(&w_s[r][half + v8 * 8]) = wreg[v8];
  }
  __syncthreads();

  // ---- phase B: per-head O contribution. wave w covers n-local
  // [w*32, w*32+32); lane l: n_local = w*32 + (l&31), b parity = l>>5.
  const int wv = tid >> 6;
  const int lane = tid & 63;
  const int nl = wv * 32 + (lane & 31);
  const int bpar = lane >> 5;                     // 0: even b, 1: odd b
  float acc[(BN + 1) / 2];
  #pragma unroll
  for (int i = 0; i < (BN + 1) / 2; ++i) acc[i] = 0.f;
  #pragma unroll 4
  for (int d = 0; d < HEAD_DIM; ++d) {
    const float wval = bf2f(w_s[nl][d]);
    #pragma unroll
    for (int i = 0; i < (BN + 1) / 2; ++i) {
      const int b = 2 * i + bpar;
      if (b < BN) acc[i] += wval * merged[b][d];
    }
  }
  #pragma unroll
  for (int i = 0; i < (BN + 1) / 2; ++i) {
    const int b = 2 * i + bpar;
    if (b < BN) atomicAdd(&o_accum[(long)b * N + n0 + nl], acc[i]);
  }
}

void attn_merge_o(torch::Tensor o_accum, torch::Tensor part,
                  torch::Tensor part_ml, torch::Tensor wo, int64_t splits) {
  const int B = o_accum.size(0), N = o_accum.size(1);
  const int n_qheads = part.size(1);
  TORCH_CHECK(B >= 1 && B <= 8, "attn_merge_o handles B<=8 (decode)");
  TORCH_CHECK(o_accum.dtype() == torch::kFloat32 && part.dtype() == torch::kFloat32);
  TORCH_CHECK(wo.dtype() == torch::kBFloat16);
  TORCH_CHECK(part.size(2) == NSPLITS_MAX && part.size(3) == HEAD_DIM);
  TORCH_CHECK(splits == 32 || splits == 64);
  TORCH_CHECK(N % AMO_NT == 0);
  TORCH_CHECK(wo.size(0) == N && wo.size(1) == n_qheads * HEAD_DIM);
  dim3 grid(N / AMO_NT, n_qheads), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  switch (B) {
#define AMO_CASE(BN) \
    case BN: hipLaunchKernelGGL((attn_merge_o_kernel<BN>), grid, block, 0, s, \
        o_accum.data_ptr<float>(), part.data_ptr<float>(), \
        part_ml.data_ptr<float>(), (const short*)wo.data_ptr(), \
        n_qheads, N, (int)splits); break;
    AMO_CASE(1) AMO_CASE(2) AMO_CASE(3) AMO_CASE(4)
    AMO_CASE(5) AMO_CASE(6) AMO_CASE(7) AMO_CASE(8)
#undef AMO_CASE
  }
  HIP_CHECK_KERNEL();
}

// ------------------------------------------------------------ router_addnorm
// x_out = x + delta (f32 attn accumulator, bf16-rounded like
// fused_add_rmsnorm); xn = rmsnorm(x_out)·γ; y = xn @ Wr^T (f32 logits).
// Writes x_out and xn once (block 0); every block stages xn in LDS for its
// four router dots. grid: N/4 (=32 for E=128); block 256.
template <int BN, bool DF32>
__global__ __launch_bounds__(256)
void router_addnorm_kernel(float* __restrict__ y,          // [BN, N]
                           const short* __restrict__ x,     // [BN, H]
                           const void* __restrict__ delta_, // [BN, H] f32|bf16
                           short* __restrict__ x_out,       // [BN, H]
                           short* __restrict__ xn_out,      // [BN, H]
                           const short* __restrict__ gamma, // [H]
                           const short* __restrict__ w,     // [N, H]
                           int H, int N, float eps) {
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int n = blockIdx.x * 4 + wid;

  extern __shared__ short xn_sh[];                // [BN][H] bf16
  __shared__ float ss_sh[BN];
  if (tid < BN) ss_sh[tid] = 0.f;
  __syncthreads();

  // pass A: xsum = bf16(x + delta) staged to LDS; accumulate sum-of-squares
  #pragma unroll
  for (int b = 0; b < BN; ++b) {
    float ss = 0.f;
    for (int base = tid * 8; base < H; base += 256 * 8) {
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + (long)b * H + base);
      float dv[8];
      if (DF32) {  // two f32x4 vector loads, not 8 scalar loads
        const f32x4 d0 = *reinterpret_cast<const f32x4*>(
            (const float*)delta_ + (long)b * H + base);
        const f32x4 d1 = *reinterpret_cast<const f32x4*>(
            (const float*)delta_ + (long)b * H + base + 4);
        dv[0] = d0[0]; dv[1] = d0[1]; dv[2] = d0[2]; dv[3] = d0[3];
        dv[4] = d1[0]; dv[5] = d1[1]; dv[6] = d1[2]; dv[7] = d1[3];
      } else {
        const bf16x8 db = *reinterpret_cast<const bf16x8*>(
            (const short*)delta_ + (long)b * H + base);
        #pragma unroll
        for (int j = 0; j < 8; ++j) dv[j] = bf2f(db[j]);
      }
      bf16x8 o;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf2f(xv[j]) + dv[j];
        o[j] = f2bf(v);
        v = bf2f(o[j]);                            // norm over the rounded sum
        ss += v * v;
      }
      *reinterpret_cast<bf16x8*>(&xn_sh[b * H + base]) = o;
    }
    ss = wave_reduce_sum(ss);
    if (lane == 0) atomicAdd(&ss_sh[b], ss);
  }
  __syncthreads();
  float scale[BN];
  #pragma unroll
  for (int b = 0; b < BN; ++b) scale[b] = rsqrtf(ss_sh[b] / H + eps);

  // pass B: block 0 persists the residual; all blocks scale LDS in place
  // (x_out write must read xsum before the in-place overwrite → same loop)
  #pragma unroll
  for (int b = 0; b < BN; ++b) {
    for (int base = tid * 8; base < H; base += 256 * 8) {
      bf16x8 sv = *reinterpret_cast<const bf16x8*>(&xn_sh[b * H + base]);
      if (blockIdx.x == 0)
        *reinterpret_cast<bf16x8*>(x_out + (long)b * H + base) = sv;
      bf16x8 gv = *reinterpret_cast<const bf16x8*>(gamma + base);
      bf16x8 o;
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f2bf(bf2f(sv[j]) * scale[b] * bf2f(gv[j]));
      *reinterpret_cast<bf16x8*>(&xn_sh[b * H + base]) = o;
      if (blockIdx.x == 0)
        *reinterpret_cast<bf16x8*>(xn_out + (long)b * H + base) = o;
    }
  }
  __syncthreads();

  // pass C: four 2048-wide dots per block out of LDS
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
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(&xn_sh[b * H + base]);
      #pragma unroll
      for (int j = 0; j < 8; ++j) acc[b] += wf[j] * bf2f(xv[j]);
    }
  }
  #pragma unroll
  for (int b = 0; b < BN; ++b) {
    const float r = wave_reduce_sum(acc[b]);
    if (lane == 0) y[(long)b * N + n] = r;
  }
}

void router_addnorm(torch::Tensor y, torch::Tensor x, torch::Tensor delta,
                    torch::Tensor x_out, torch::Tensor xn_out,
                    torch::Tensor gamma, torch::Tensor w, double eps) {
  const int B = x.size(0), H = x.size(1), N = w.size(0);
  TORCH_CHECK(B >= 1 && B <= 8, "router_addnorm handles B<=8 (decode)");
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && w.dtype() == torch::kBFloat16);
  const bool df32 = delta.dtype() == torch::kFloat32;
  TORCH_CHECK(df32 || delta.dtype() == torch::kBFloat16);
  TORCH_CHECK(delta.is_contiguous());
  TORCH_CHECK(y.dtype() == torch::kFloat32);
  TORCH_CHECK(H % (WAVE * 8) == 0);
  const size_t lds = (size_t)B * H * sizeof(short);
  TORCH_CHECK(lds <= 160 * 1024, "B*H too large for LDS staging");
  dim3 grid((N + 3) / 4), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  switch (B * 2 + (df32 ? 1 : 0)) {
#define RAN_CASE1(BN, DF) \
    case BN * 2 + (DF ? 1 : 0): \
      hipLaunchKernelGGL((router_addnorm_kernel<BN, DF>), grid, block, lds, s, \
        y.data_ptr<float>(), (const short*)x.data_ptr(), \
        delta.data_ptr(), (short*)x_out.data_ptr(), \
        (short*)xn_out.data_ptr(), (const short*)gamma.data_ptr(), \
        (const short*)w.data_ptr(), H, N, (float)eps); break;
#define RAN_CASE(BN) RAN_CASE1(BN, true) RAN_CASE1(BN, false)
    RAN_CASE(1) RAN_CASE(2) RAN_CASE(3) RAN_CASE(4)
    RAN_CASE(5) RAN_CASE(6) RAN_CASE(7) RAN_CASE(8)
#undef RAN_CASE
#undef RAN_CASE1
  }
  HIP_CHECK_KERNEL();
}
