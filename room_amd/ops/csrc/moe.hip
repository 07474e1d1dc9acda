// Mixture-of-Experts kernels for Qwen3-30B-A3B on CDNA4 (gfx950).
//   E=128 experts, top-8, hidden H=2048, expert intermediate I=768.
//
// Weight layouts (we own them):
//   W13: [E, 2*I, H]  — gate rows [0,I), up rows [I,2I); each row contiguous
//                       over H (GEMV-friendly streaming, MFMA B-operand rows)
//   W2:  [E, H, I]    — down-proj rows contiguous over I
//
// Decode path (few tokens): wave-per-output GEMV — each (token,expert) pair
//   is bandwidth-bound on expert weights; 64-lane dot with 16 B/lane loads.
// Prefill path (many tokens): grouped MFMA GEMM (16x16x32 bf16) with LDS
//   tiles and +16B row padding against bank conflicts; tokens pre-sorted by
//   expert on host, tile descriptors built host-side.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

// ---------------------------------------------------------------- router
// softmax over E=128 logits, pick top-K=8, renormalize their probs.
// one wave per token; lane l owns logits[2l], [2l+1].
__global__ void moe_router_kernel(int* __restrict__ topk_ids,      // [T, K]
                                  float* __restrict__ topk_w,      // [T, K]
                                  const float* __restrict__ logits,  // [T, E]
                                  int E, int K) {
  const int t = blockIdx.x;
  const int lane = threadIdx.x;
  const float l0 = (2 * lane < E) ? logits[(long)t * E + 2 * lane] : -INFINITY;
  const float l1 = (2 * lane + 1 < E) ? logits[(long)t * E + 2 * lane + 1] : -INFINITY;

  float m = wave_reduce_max(fmaxf(l0, l1));
  float e0 = __expf(l0 - m), e1 = __expf(l1 - m);
  float denom = wave_reduce_sum(e0 + e1);

  // iterative top-K selection by masking (K ≤ 8)
  float v0 = l0, v1 = l1;
  float picked_sum = 0.f;
  float probs[8];
  int winners[8];
  for (int k = 0; k < K; ++k) {
    float best = fmaxf(v0, v1);
    float gmax = wave_reduce_max(best);
    // first lane holding gmax wins; prefer slot 0
    int cand_id = (v0 == gmax) ? 2 * lane : ((v1 == gmax) ? 2 * lane + 1 : INT_MAX);
    int winner = cand_id;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      winner = min(winner, __shfl_xor(winner, off, WAVE));
    float prob = __expf(gmax - m) / denom;
    probs[k] = prob;       // every lane tracks the same values
    winners[k] = winner;
    picked_sum += prob;
    if (winner == 2 * lane) v0 = -INFINITY;
    if (winner == 2 * lane + 1) v1 = -INFINITY;
  }
  // lane 0 writes ids + renormalized weights (Qwen3 norm_topk_prob) — single
  // writer, no cross-lane visibility assumptions on global memory
  if (lane == 0) {
    for (int k = 0; k < K; ++k) {
      topk_ids[(long)t * K + k] = winners[k];
      topk_w[(long)t * K + k] = probs[k] / picked_sum;
    }
  }
}

// ---------------------------------------------------------------- fused router
// Decode-path router in ONE kernel: logits = x·Wr^T, softmax over E, top-K,
// renormalize. Replaces a dense GEMV launch + moe_router launch + an f32
// logits round-trip (~18 µs/layer incl. bubbles at B=5 → ~3 µs).
// Grid: T blocks × 256 threads (4 waves). Wave w computes logits
// [w*32, w*32+32) by coalesced W-row reads against an LDS-staged x; then
// wave 0 runs the same masking top-K as moe_router_kernel from LDS logits.
__global__ __launch_bounds__(256)
void router_topk_kernel(int* __restrict__ topk_ids,     // [T, K]
                        float* __restrict__ topk_w,     // [T, K]
                        const short* __restrict__ x,    // [T, H] bf16
                        const short* __restrict__ wr,   // [E, H] bf16
                        int H, int E, int K) {
  const int t = blockIdx.x;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;

  extern __shared__ short smem[];
  short* x_s = smem;                                  // [H]
  float* l_s = reinterpret_cast<float*>(smem + ((H + 15) & ~15));  // [E]

  for (int i = tid * 8; i < H; i += blockDim.x * 8)
    *reinterpret_cast<bf16x8*>(&x_s[i]) =
        *reinterpret_cast<const bf16x8*>(x + (long)t * H + i);
  __syncthreads();

  const int per_wave = (E + 3) / 4;                   // logits per wave
  for (int i = 0; i < per_wave; ++i) {
    const int e = wid * per_wave + i;
    if (e >= E) break;
    const short* wrow = wr + (long)e * H;
    float dot = 0.f;
    for (int base = lane * 8; base < H; base += WAVE * 8) {
      bf16x8 wv = *reinterpret_cast<const bf16x8*>(wrow + base);
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(&x_s[base]);
      #pragma unroll
      for (int j = 0; j < 8; ++j) dot += bf2f(wv[j]) * bf2f(xv[j]);
    }
    dot = wave_reduce_sum(dot);
    if (lane == 0) l_s[e] = dot;
  }
  __syncthreads();
  if (wid != 0) return;

  // softmax + masking top-K over LDS logits (same algorithm as moe_router)
  const float l0 = (2 * lane < E) ? l_s[2 * lane] : -INFINITY;
  const float l1 = (2 * lane + 1 < E) ? l_s[2 * lane + 1] : -INFINITY;
  float m = wave_reduce_max(fmaxf(l0, l1));
  float denom = wave_reduce_sum(__expf(l0 - m) + __expf(l1 - m));
  float v0 = l0, v1 = l1;
  float picked_sum = 0.f;
  float probs[8];
  int winners[8];
  for (int k = 0; k < K; ++k) {
    float gmax = wave_reduce_max(fmaxf(v0, v1));
    int cand_id = (v0 == gmax) ? 2 * lane : ((v1 == gmax) ? 2 * lane + 1 : INT_MAX);
    int winner = cand_id;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      winner = min(winner, __shfl_xor(winner, off, WAVE));
    probs[k] = __expf(gmax - m) / denom;
    winners[k] = winner;
    picked_sum += probs[k];
    if (winner == 2 * lane) v0 = -INFINITY;
    if (winner == 2 * lane + 1) v1 = -INFINITY;
  }
  if (lane == 0) {
    for (int k = 0; k < K; ++k) {
      topk_ids[(long)t * K + k] = winners[k];
      topk_w[(long)t * K + k] = probs[k] / picked_sum;
    }
  }
}

// ------------------------------------------------- H-split router GEMV
// The router projection (N=128 outputs) maps to only N/4=32 workgroups in
// the generic wave-per-output GEMV — latency-starved at 10.8 µs for 0.5 MB.
// Split H into RS=4 slices: wave (n, s) computes a partial dot (4× the
// parallelism, one 512-element burst per wave); the router top-k kernel
// sums the 4 partials when it loads each logit (an f32x4 vector read).
#define RS 4

__global__ __launch_bounds__(256)
void router_gemv_partial_kernel(float* __restrict__ yp,   // [B, N, RS]
                                const short* __restrict__ x,   // [B, H]
                                const short* __restrict__ w,   // [N, H]
                                int B, int H, int N) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = blockIdx.x * 4 + wid;
  const int s = blockIdx.y;
  if (n >= N) return;
  const int hs = H / RS;
  const int h0 = s * hs;
  float acc[8];
  #pragma unroll
  for (int b = 0; b < 8; ++b) acc[b] = 0.f;
  const short* wrow = w + (long)n * H + h0;
  for (int base = lane * 8; base < hs; base += WAVE * 8) {
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(wrow + base);
    float wf[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) wf[j] = bf2f(wv[j]);
    #pragma unroll
    for (int b = 0; b < 8; ++b) {
      if (b < B) {
        bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + (long)b * H + h0 + base);
        #pragma unroll
        for (int j = 0; j < 8; ++j) acc[b] += wf[j] * bf2f(xv[j]);
      }
    }
  }
  #pragma unroll
  for (int b = 0; b < 8; ++b) {
    if (b < B) {
      const float r = wave_reduce_sum(acc[b]);
      if (lane == 0) yp[((long)b * N + n) * RS + s] = r;
    }
  }
}

// moe_router over RS-partial logits: lane l owns logits 2l, 2l+1; each is an
// f32x4 load + horizontal sum. Identical selection/renorm to moe_router.
__global__ void moe_router_p4_kernel(int* __restrict__ topk_ids,
                                     float* __restrict__ topk_w,
                                     const float* __restrict__ yp,  // [T, E, RS]
                                     int E, int K) {
  const int t = blockIdx.x;
  const int lane = threadIdx.x;
  float l0 = -INFINITY, l1 = -INFINITY;
  if (2 * lane < E) {
    const f32x4 v = *reinterpret_cast<const f32x4*>(
        yp + ((long)t * E + 2 * lane) * RS);
    l0 = v[0] + v[1] + v[2] + v[3];
  }
  if (2 * lane + 1 < E) {
    const f32x4 v = *reinterpret_cast<const f32x4*>(
        yp + ((long)t * E + 2 * lane + 1) * RS);
    l1 = v[0] + v[1] + v[2] + v[3];
  }

  float m = wave_reduce_max(fmaxf(l0, l1));
  float denom = wave_reduce_sum(__expf(l0 - m) + __expf(l1 - m));
  float v0 = l0, v1 = l1;
  float picked_sum = 0.f;
  float probs[8];
  int winners[8];
  for (int k = 0; k < K; ++k) {
    float gmax = wave_reduce_max(fmaxf(v0, v1));
    int cand_id = (v0 == gmax) ? 2 * lane : ((v1 == gmax) ? 2 * lane + 1 : INT_MAX);
    int winner = cand_id;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      winner = min(winner, __shfl_xor(winner, off, WAVE));
    probs[k] = __expf(gmax - m) / denom;
    winners[k] = winner;
    picked_sum += probs[k];
    if (winner == 2 * lane) v0 = -INFINITY;
    if (winner == 2 * lane + 1) v1 = -INFINITY;
  }
  if (lane == 0) {
    for (int k = 0; k < K; ++k) {
      topk_ids[(long)t * K + k] = winners[k];
      topk_w[(long)t * K + k] = probs[k] / picked_sum;
    }
  }
}

// ---------------------------------------------------------------- GEMV path
// h[p][j] = silu(dot(x_t, Wg_row_j)) * dot(x_t, Wu_row_j)
// grid: (npairs, I/4); block 256 = 4 waves; wave w computes output j.
__global__ __launch_bounds__(256)
void moe_gemv_h_kernel(short* __restrict__ h,             // [P, I]
                       const short* __restrict__ x,        // [T, H]
                       const short* __restrict__ w13,      // [E, 2I, H]
                       const int* __restrict__ pair_token,  // [P]
                       const int* __restrict__ pair_expert, // [P]
                       float* __restrict__ out_zero,        // [zero_n] or null
                       long zero_n, int H, int I) {
  // side job: zero the downstream accumulator (out of moe_gemv_down) here —
  // the grid has ~2M threads vs ~10k floats to clear, and this kernel always
  // runs right before the down kernel, so the separate fill launch it
  // replaces (4.8 µs/layer, 2.8% of the decode step) is absorbed for free.
  if (out_zero != nullptr) {
    const long gid = ((long)blockIdx.y * gridDim.x + blockIdx.x) * blockDim.x
                     + threadIdx.x;
    if (gid < zero_n) out_zero[gid] = 0.f;
  }
  const int p = blockIdx.x;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int j = blockIdx.y * 4 + wid;
  if (j >= I) return;
  const int t = pair_token[p];
  const int e = pair_expert[p];
  const short* xrow = x + (long)t * H;
  const short* grow = w13 + ((long)e * 2 * I + j) * H;
  const short* urow = w13 + ((long)e * 2 * I + I + j) * H;

  float dg = 0.f, du = 0.f;
  for (int base = lane * 8; base < H; base += WAVE * 8) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(xrow + base);
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(grow + base);
    bf16x8 uv = *reinterpret_cast<const bf16x8*>(urow + base);
    #pragma unroll
    for (int q_ = 0; q_ < 8; ++q_) {
      float xf = bf2f(xv[q_]);
      dg += xf * bf2f(gv[q_]);
      du += xf * bf2f(uv[q_]);
    }
  }
  dg = wave_reduce_sum(dg);
  du = wave_reduce_sum(du);
  if (lane == 0) {
    float s = dg / (1.0f + __expf(-dg));
    h[(long)p * I + j] = f2bf(s * du);
  }
}

// z[t] += w_p * (h_p @ W2_e^T): wave per output dim o; atomic f32 add.
__global__ __launch_bounds__(256)
void moe_gemv_down_kernel(float* __restrict__ out,         // [T, H] f32
                          const short* __restrict__ h,      // [P, I]
                          const short* __restrict__ w2,     // [E, H, I]
                          const float* __restrict__ pair_w,  // [P]
                          const int* __restrict__ pair_token,
                          const int* __restrict__ pair_expert,
                          int H, int I) {
  // 16 lanes per output, 4 outputs per wave: the I=768 row is 1.5 full-wave
  // bursts (the second one 50% idle) and a 6-shuffle reduction per 1.5 KB in
  // the wave-per-output shape; 16-lane groups keep every lane busy (6 bursts
  // of 128 elems) and cut the reduction to 4 shuffles per output.
  const int p = blockIdx.x;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int grp = lane >> 4;       // output within the wave
  const int sub = lane & 15;       // 16 lanes per output
  const int o = (blockIdx.y * 4 + wid) * 4 + grp;
  if (o >= H) return;
  const int t = pair_token[p];
  const int e = pair_expert[p];
  const short* hrow = h + (long)p * I;
  const short* wrow = w2 + ((long)e * H + o) * I;

  float d = 0.f;
  for (int base = sub * 8; base < I; base += 16 * 8) {
    bf16x8 hv = *reinterpret_cast<const bf16x8*>(hrow + base);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(wrow + base);
    #pragma unroll
    for (int q_ = 0; q_ < 8; ++q_) d += bf2f(hv[q_]) * bf2f(wv[q_]);
  }
  #pragma unroll
  for (int off = 8; off > 0; off >>= 1) d += __shfl_xor(d, off, WAVE);
  if (sub == 0)
    atomicAdd(out + (long)t * H + o, d * pair_w[p]);
}

// ------------------------------------------------ grouped MFMA GEMM (prefill)
// C[P_sorted, N] = X[token(p), :] @ W[e, :, :]^T for tokens grouped by expert.
// Tile: BM=16 rows (pairs), BN=64 (4 waves × 16), BK=64.
// tile_desc per workgroup: (expert, pair_row_start, m_size, n_tile)
//
// MFMA fragment layouts (gfx950 mfma_f32_16x16x32_bf16, verified vs torch in
// tests/test_kernels_gpu.py):
//   A (16×32): lane l holds A[l&15][(l>>4)*8 .. +8]
//   B (32×16): lane l holds B[(l>>4)*8 .. +8][l&15]
//   C (16×16): lane l, reg r ↦ row (l>>4)*4 + r, col l&15
typedef __attribute__((ext_vector_type(4))) float cfrag_t;

#define GG_BM 16
#define GG_BN 64
#define GG_BK 64
#define GG_PAD 8   // bf16 elements of row padding in LDS (16B, keeps b128 alignment)

__global__ __launch_bounds__(256)
void moe_grouped_gemm_kernel(short* __restrict__ out,       // [P, N] bf16
                             const short* __restrict__ x,    // [T, H]
                             const short* __restrict__ w,    // [E, N, H]
                             const int* __restrict__ pair_token,   // [P] sorted
                             const int* __restrict__ tile_desc,    // [G, 4]
                             int H, int N) {
  const int g = blockIdx.x;
  const int e = tile_desc[g * 4 + 0];
  const int row0 = tile_desc[g * 4 + 1];
  const int msize = tile_desc[g * 4 + 2];
  const int n0 = tile_desc[g * 4 + 3] * GG_BN;

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;

  __shared__ short xs[GG_BM][GG_BK + GG_PAD];
  __shared__ short ws[GG_BN][GG_BK + GG_PAD];

  const short* wbase = w + (long)e * N * H;

  cfrag_t acc = {0.f, 0.f, 0.f, 0.f};  // each wave: one 16×16 C fragment at n0+wid*16

  for (int k0 = 0; k0 < H; k0 += GG_BK) {
    // stage X tile: 16 rows × 64 k = 128 × 8-vec; 256 threads → first 128
    {
      const int nvec = GG_BM * GG_BK / 8;  // 128
      if (tid < nvec) {
        const int r = (tid * 8) / GG_BK;
        const int c = (tid * 8) % GG_BK;
        bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (r < msize) {
          const int tok = pair_token[row0 + r];
          v = *reinterpret_cast<const bf16x8*>(x + (long)tok * H + k0 + c);
        }
        *reinterpret_cast<bf16x8*>(&xs[r][c]) = v;
      }
    }
    // stage W tile: 64 rows × 64 k = 512 × 8-vec; 256 threads → 2 each
    {
      #pragma unroll
      for (int it = 0; it < 2; ++it) {
        const int idx = tid + it * 256;
        const int r = (idx * 8) / GG_BK;
        const int c = (idx * 8) % GG_BK;
        bf16x8 v = *reinterpret_cast<const bf16x8*>(
            wbase + (long)(n0 + r) * H + k0 + c);
        *reinterpret_cast<bf16x8*>(&ws[r][c]) = v;
      }
    }
    __syncthreads();

    // 2 MFMA per K-tile (K=32 each)
    #pragma unroll
    for (int kk = 0; kk < GG_BK / 32; ++kk) {
      const int arow = lane & 15;
      const int akoff = (lane >> 4) * 8;
      bf16x8 a = *reinterpret_cast<const bf16x8*>(&xs[arow][kk * 32 + akoff]);
      const int bcol = wid * 16 + (lane & 15);
      bf16x8 b = *reinterpret_cast<const bf16x8*>(&ws[bcol][kk * 32 + akoff]);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: C fragment — lane l, reg r → row (l>>4)*4+r, col l&15
  const int crow = (lane >> 4) * 4;
  const int ccol = wid * 16 + (lane & 15);
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = crow + r;
    if (m < msize && n0 + ccol < N)
      out[(long)(row0 + m) * N + n0 + ccol] = f2bf(acc[r]);
  }
}

// combine: out[t] += w_p * z[p]  (vectorized scatter-add over sorted pairs)
__global__ void moe_combine_kernel(float* __restrict__ out,     // [T, H] f32
                                   const short* __restrict__ z,  // [P, H]
                                   const float* __restrict__ pair_w,
                                   const int* __restrict__ pair_token,
                                   int H) {
  const int p = blockIdx.x;
  const int t = pair_token[p];
  const float wp = pair_w[p];
  for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
    bf16x8 zv = *reinterpret_cast<const bf16x8*>(z + (long)p * H + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      atomicAdd(out + (long)t * H + i + j, wp * bf2f(zv[j]));
  }
}

// ---------------------------------------------------------------- wrappers

void moe_router(torch::Tensor topk_ids, torch::Tensor topk_w, torch::Tensor logits,
                int64_t K) {
  const int T = logits.size(0), E = logits.size(1);
  TORCH_CHECK(E <= 128, "router kernel handles E<=128");
  TORCH_CHECK(logits.dtype() == torch::kFloat32);
  dim3 grid(T), block(WAVE);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(moe_router_kernel, grid, block, 0, s,
                     topk_ids.data_ptr<int>(), topk_w.data_ptr<float>(),
                     logits.data_ptr<float>(), E, (int)K);
  HIP_CHECK_KERNEL();
}

void router_gemv_topk(torch::Tensor topk_ids, torch::Tensor topk_w,
                      torch::Tensor x, torch::Tensor wr, int64_t K) {
  const int B = x.size(0), H = x.size(1), E = wr.size(0);
  TORCH_CHECK(E <= 128 && K <= 8 && B <= 8);
  TORCH_CHECK(H % (RS * WAVE * 8) == 0 || (H / RS) % 8 == 0,
              "H/RS must be a multiple of 8");
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && wr.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && wr.is_contiguous());
  // persistent partial buffer per (B, E): allocation-free on graph replay
  static std::unordered_map<long, torch::Tensor> parts;
  const long key = (long)B * 1024 + E;
  auto it = parts.find(key);
  if (it == parts.end())
    it = parts.emplace(key, torch::empty(
        {(long)B * E * RS},
        torch::TensorOptions().device(x.device()).dtype(torch::kFloat32))).first;
  hipStream_t s = c10::hip::getCurrentHIPStream();
  dim3 g1((E + 3) / 4, RS);
  hipLaunchKernelGGL(router_gemv_partial_kernel, g1, dim3(256), 0, s,
                     it->second.data_ptr<float>(), (const short*)x.data_ptr(),
                     (const short*)wr.data_ptr(), B, H, E);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(moe_router_p4_kernel, dim3(B), dim3(WAVE), 0, s,
                     topk_ids.data_ptr<int>(), topk_w.data_ptr<float>(),
                     it->second.data_ptr<float>(), E, (int)K);
  HIP_CHECK_KERNEL();
}

void router_topk(torch::Tensor topk_ids, torch::Tensor topk_w, torch::Tensor x,
                 torch::Tensor wr, int64_t K) {
  const int T = x.size(0), H = x.size(1), E = wr.size(0);
  TORCH_CHECK(E <= 128 && K <= 8, "router kernel handles E<=128, K<=8");
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && wr.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && wr.is_contiguous());
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  const size_t lds = ((H + 15) & ~15) * sizeof(short) + E * sizeof(float);
  dim3 grid(T), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(router_topk_kernel, grid, block, lds, s,
                     topk_ids.data_ptr<int>(), topk_w.data_ptr<float>(),
                     (const short*)x.data_ptr(), (const short*)wr.data_ptr(),
                     H, E, (int)K);
  HIP_CHECK_KERNEL();
}

void moe_gemv_h(torch::Tensor h, torch::Tensor x, torch::Tensor w13,
                torch::Tensor pair_token, torch::Tensor pair_expert,
                torch::Tensor out_zero) {
  const int P = pair_token.size(0);
  const int H = x.size(-1), I = h.size(-1);
  dim3 grid(P, (I + 3) / 4), block(256);
  float* zp = nullptr;
  long zn = 0;
  if (out_zero.defined() && out_zero.numel() > 0) {
    TORCH_CHECK(out_zero.dtype() == torch::kFloat32);
    zp = out_zero.data_ptr<float>();
    zn = out_zero.numel();
    TORCH_CHECK(zn <= (long)grid.x * grid.y * block.x,
                "zero target larger than grid coverage");
  }
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(moe_gemv_h_kernel, grid, block, 0, s,
                     (short*)h.data_ptr(), (const short*)x.data_ptr(),
                     (const short*)w13.data_ptr(), pair_token.data_ptr<int>(),
                     pair_expert.data_ptr<int>(), zp, zn, H, I);
  HIP_CHECK_KERNEL();
}

void moe_gemv_down(torch::Tensor out, torch::Tensor h, torch::Tensor w2,
                   torch::Tensor pair_w, torch::Tensor pair_token,
                   torch::Tensor pair_expert) {
  const int P = pair_token.size(0);
  const int H = out.size(-1), I = h.size(-1);
  TORCH_CHECK(out.dtype() == torch::kFloat32);
  TORCH_CHECK(I % (16 * 8) == 0, "I must be a multiple of 128");
  dim3 grid(P, (H + 15) / 16), block(256);   // 16 outputs per WG
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(moe_gemv_down_kernel, grid, block, 0, s,
                     out.data_ptr<float>(), (const short*)h.data_ptr(),
                     (const short*)w2.data_ptr(), pair_w.data_ptr<float>(),
                     pair_token.data_ptr<int>(), pair_expert.data_ptr<int>(), H, I);
  HIP_CHECK_KERNEL();
}

void moe_grouped_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                      torch::Tensor pair_token, torch::Tensor tile_desc) {
  const int H = x.size(-1), N = out.size(-1);
  const int G = tile_desc.size(0);
  TORCH_CHECK(H % GG_BK == 0 && N % GG_BN == 0);
  dim3 grid(G), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(moe_grouped_gemm_kernel, grid, block, 0, s,
                     (short*)out.data_ptr(), (const short*)x.data_ptr(),
                     (const short*)w.data_ptr(), pair_token.data_ptr<int>(),
                     tile_desc.data_ptr<int>(), H, N);
  HIP_CHECK_KERNEL();
}

void moe_combine(torch::Tensor out, torch::Tensor z, torch::Tensor pair_w,
                 torch::Tensor pair_token) {
  const int P = pair_token.size(0), H = out.size(-1);
  TORCH_CHECK(out.dtype() == torch::kFloat32);
  dim3 grid(P), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(moe_combine_kernel, grid, block, 0, s,
                     out.data_ptr<float>(), (const short*)z.data_ptr(),
                     pair_w.data_ptr<float>(), pair_token.data_ptr<int>(), H);
  HIP_CHECK_KERNEL();
}

// ------------------------------------------------ BM=128 grouped MFMA GEMM
// The BM=16 tile re-reads each expert's weight panel ceil(rows/16)× per layer
// (~8× at 2k-token prefill → 6.4 GB/layer). BM=128 reads W once per expert
// m-tile: 8 waves (512 threads), wave w owns rows [16w, 16w+16) × BN=64.
// LDS rows padded +8 bf16 so ds_read_b128 across rows spreads banks (G4).

#define G2_BM 128
#define G2_BN 64
#define G2_BK 64
#define G2_PAD 8

// BN templated: wider N-tiles cut the X-tile re-read factor (X is re-read
// once per N-tile; gateup N=1536 at BN=64 → 24×, at BN=128 → 12×).
template <int BM, int BN, int BK = G2_BK>
__global__ __launch_bounds__(512)
void moe_grouped_gemm128_kernel(short* __restrict__ out,      // [P, N]
                                const short* __restrict__ x,   // [T, H]
                                const short* __restrict__ w,   // [E, N, H]
                                const int* __restrict__ pair_token,
                                const int* __restrict__ tile_desc,  // [G,3]
                                int H, int N) {
  const int g = blockIdx.x;
  const int msize = tile_desc[g * 3 + 2];
  if (msize == 0) return;                 // past the live tile count
  const int e = tile_desc[g * 3 + 0];
  const int row0 = tile_desc[g * 3 + 1];
  const int n0 = blockIdx.y * BN;         // n-tile from the grid

  const int tid = threadIdx.x;
  const int wid = tid >> 6;       // wave index: MF 16-row m-slices each
  const int lane = tid & 63;

  __shared__ short xs[BM][BK + G2_PAD];
  __shared__ short ws[BN][BK + G2_PAD];

  const short* wbase = w + (long)e * N * H;
  constexpr int NF = BN / 16;             // n-fragments per wave
  constexpr int MF = BM / 128;            // m-fragments per wave (1 or 2)

  cfrag_t acc[MF][NF];
  #pragma unroll
  for (int mf = 0; mf < MF; ++mf)
    #pragma unroll
    for (int nf = 0; nf < NF; ++nf) acc[mf][nf] = cfrag_t{0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < H; k0 += BK) {
    // stage X tile: BM rows × 64 k = BM*8 vec8 → BM/64 per thread
    #pragma unroll
    for (int it = 0; it < BM * BK / 64 / G2_BK; ++it) {
      const int idx = tid + it * 512;
      const int r = (idx * 8) / BK;
      const int c = (idx * 8) % BK;
      bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (r < msize) {
        const int tok = pair_token[row0 + r];
        v = *reinterpret_cast<const bf16x8*>(x + (long)tok * H + k0 + c);
      }
      *reinterpret_cast<bf16x8*>(&xs[r][c]) = v;
    }
    // stage W tile: BN rows × 64 k → BN/64 vec8 per thread
    #pragma unroll
    for (int it = 0; it < BN * BK / 64 / G2_BK; ++it) {
      const int idx = tid + it * 512;
      const int r = (idx * 8) / BK;
      const int c = (idx * 8) % BK;
      bf16x8 v = *reinterpret_cast<const bf16x8*>(
          wbase + (long)(n0 + r) * H + k0 + c);
      *reinterpret_cast<bf16x8*>(&ws[r][c]) = v;
    }
    __syncthreads();

    #pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      const int akoff = kk * 32 + (lane >> 4) * 8;
      #pragma unroll
      for (int mf = 0; mf < MF; ++mf) {
        const int arow = mf * 128 + wid * 16 + (lane & 15);
        bf16x8 a = *reinterpret_cast<const bf16x8*>(&xs[arow][akoff]);
        #pragma unroll
        for (int nf = 0; nf < NF; ++nf) {
          const int bcol = nf * 16 + (lane & 15);
          bf16x8 b = *reinterpret_cast<const bf16x8*>(&ws[bcol][akoff]);
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[mf][nf], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  const int ccol_base = lane & 15;
  #pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int crow = mf * 128 + wid * 16 + (lane >> 4) * 4;
    #pragma unroll
    for (int nf = 0; nf < NF; ++nf) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = crow + r;
        if (m < msize)
          out[(long)(row0 + m) * N + n0 + nf * 16 + ccol_base] = f2bf(acc[mf][nf][r]);
      }
    }
  }
}

void moe_grouped_gemm128(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                         torch::Tensor pair_token, torch::Tensor tile_desc,
                         int64_t bm) {
  const int H = x.size(-1), N = out.size(-1);
  const int G = tile_desc.size(0);
  TORCH_CHECK(H % G2_BK == 0 && N % G2_BN == 0);
  TORCH_CHECK(tile_desc.size(1) == 3, "desc is [G,3]; n-tile comes from grid.y");
  hipStream_t s = c10::hip::getCurrentHIPStream();
  // measured: BN=128 is ~11% SLOWER at 4k-token prefill (10.3k vs 11.6k
  // tok/s) — L2 already absorbs the X-tile re-reads and wider tiles halve
  // the block count (less latency hiding). BN=64 default; 128 kept for A/B.
  static const bool wide = []() {
    const char* v = getenv("ROOMAMD_MOE_BN128");
    return v && v[0] == '1';
  }();
  const int BM = (int)bm;
  TORCH_CHECK(BM == 128 || BM == 256, "bm must be 128 or 256");
  if (wide && N % 128 == 0) {
    dim3 grid(G, N / 128), block(512);
    if (BM == 256)
      hipLaunchKernelGGL((moe_grouped_gemm128_kernel<256, 128>), grid, block, 0, s,
                         (short*)out.data_ptr(), (const short*)x.data_ptr(),
                         (const short*)w.data_ptr(), pair_token.data_ptr<int>(),
                         tile_desc.data_ptr<int>(), H, N);
    else
      hipLaunchKernelGGL((moe_grouped_gemm128_kernel<128, 128>), grid, block, 0, s,
                         (short*)out.data_ptr(), (const short*)x.data_ptr(),
                         (const short*)w.data_ptr(), pair_token.data_ptr<int>(),
                         tile_desc.data_ptr<int>(), H, N);
  } else {
    dim3 grid(G, N / G2_BN), block(512);
    static const bool bk128 = []() {
      const char* v = getenv("ROOMAMD_MOE_BK128");
      return v && v[0] == '1';
    }();
    if (BM == 256)
      hipLaunchKernelGGL((moe_grouped_gemm128_kernel<256, G2_BN>), grid, block, 0, s,
                         (short*)out.data_ptr(), (const short*)x.data_ptr(),
                         (const short*)w.data_ptr(), pair_token.data_ptr<int>(),
                         tile_desc.data_ptr<int>(), H, N);
    else if (bk128 && H % 128 == 0)
      hipLaunchKernelGGL((moe_grouped_gemm128_kernel<128, G2_BN, 128>), grid, block, 0, s,
                         (short*)out.data_ptr(), (const short*)x.data_ptr(),
                         (const short*)w.data_ptr(), pair_token.data_ptr<int>(),
                         tile_desc.data_ptr<int>(), H, N);
    else
      hipLaunchKernelGGL((moe_grouped_gemm128_kernel<128, G2_BN>), grid, block, 0, s,
                         (short*)out.data_ptr(), (const short*)x.data_ptr(),
                         (const short*)w.data_ptr(), pair_token.data_ptr<int>(),
                         tile_desc.data_ptr<int>(), H, N);
  }
  HIP_CHECK_KERNEL();
}

// gather-combine: out[t] = Σ_k w[t,k] * z[inv_order[t*K+k]] — replaces the
// atomicAdd combine (33M atomics/layer at 2k-token prefill → 0.9 ms/layer).
// inv_order maps (token, k) → sorted-pair row of z.
__global__ void moe_combine_gather_kernel(short* __restrict__ out,  // [T, H] bf16
                                          const short* __restrict__ z,  // [P, H]
                                          const float* __restrict__ topk_w,  // [T, K]
                                          const int* __restrict__ inv_order,  // [T*K]
                                          int H, int K) {
  const int t = blockIdx.x;
  float w[8];
  int p[8];
  for (int k = 0; k < K; ++k) {
    w[k] = topk_w[(long)t * K + k];
    p[k] = inv_order[(long)t * K + k];
  }
  for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int k = 0; k < K; ++k) {
      bf16x8 zv = *reinterpret_cast<const bf16x8*>(z + (long)p[k] * H + i);
      #pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += w[k] * bf2f(zv[j]);
    }
    bf16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(acc[j]);
    *reinterpret_cast<bf16x8*>(out + (long)t * H + i) = o;
  }
}

void moe_combine_gather(torch::Tensor out, torch::Tensor z, torch::Tensor topk_w,
                        torch::Tensor inv_order) {
  const int T = out.size(0), H = out.size(-1);
  const int K = topk_w.size(-1);
  TORCH_CHECK(K <= 8 && out.dtype() == torch::kBFloat16);
  dim3 grid(T), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(moe_combine_gather_kernel, grid, block, 0, s,
                     (short*)out.data_ptr(), (const short*)z.data_ptr(),
                     topk_w.data_ptr<float>(), inv_order.data_ptr<int>(), H, K);
  HIP_CHECK_KERNEL();
}

// device-side tile-descriptor builder: removes the per-layer host syncs that
// data-dependent repeat_interleave shapes forced (profiled: ~6 syncs/layer
// serialized prefill on the CPU). desc[GMAX,3] = (expert, row_start, m_size);
// unused slots get m_size=0 and the GEMM early-exits. GMAX is host-computed
// from P alone (≤ E + ceil(P/bm)), so grids stay static.
__global__ void moe_build_desc_kernel(int* __restrict__ desc,      // [GMAX, 3]
                                      const long* __restrict__ counts,  // [E]
                                      int E, int bm, int gmax) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  int g = 0;
  long row = 0;
  for (int e = 0; e < E; ++e) {
    const long c = counts[e];
    for (long m0 = 0; m0 < c && g < gmax; m0 += bm) {
      desc[g * 3 + 0] = e;
      desc[g * 3 + 1] = (int)(row + m0);
      desc[g * 3 + 2] = (int)min((long)bm, c - m0);
      ++g;
    }
    row += c;
  }
  for (; g < gmax; ++g) desc[g * 3 + 2] = 0;
}

void moe_build_desc(torch::Tensor desc, torch::Tensor counts, int64_t bm) {
  const int E = counts.numel();
  const int gmax = desc.size(0);
  TORCH_CHECK(counts.dtype() == torch::kInt64 && desc.size(1) == 3);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(moe_build_desc_kernel, dim3(1), dim3(64), 0, s,
                     desc.data_ptr<int>(), counts.data_ptr<long>(), E, (int)bm,
                     gmax);
  HIP_CHECK_KERNEL();
}

// ------------------------------------------------ deduped decode GEMV (v2)
// The pair-GEMV reads an expert's gate/up/down rows once per (token,expert)
// pair; with B×K pairs over 128 experts many experts serve several tokens
// (B=5: ~40 pairs → ~35 unique experts). v2 groups pairs by expert with a
// tiny list-building kernel (static [E, CAP] shapes → hipGraph-safe), then
// each weight row is read ONCE and applied to all of that expert's tokens
// (x/h rows are L2-resident at decode sizes).

#define MD_CAP 64   // max pairs per expert per step (B≤8 × K=8)

__global__ void moe_build_lists_kernel(int* __restrict__ counts,   // [E]
                                       int* __restrict__ tok_list,  // [E, CAP]
                                       float* __restrict__ w_list,  // [E, CAP]
                                       int* __restrict__ active,    // [P] compacted
                                       const int* __restrict__ topk_ids,  // [T,K]
                                       const float* __restrict__ topk_w,
                                       int T, int K, int E) {
  // single block: E ≤ 128 and T*K ≤ 512 at decode — µs-scale
  if (blockIdx.x != 0) return;
  for (int e = threadIdx.x; e < E; e += blockDim.x) counts[e] = 0;
  __syncthreads();
  for (int p = threadIdx.x; p < T * K; p += blockDim.x) {
    const int e = topk_ids[p];
    const int slot = atomicAdd(&counts[e], 1);
    if (slot < MD_CAP) {
      tok_list[e * MD_CAP + slot] = p / K;   // token index
      w_list[e * MD_CAP + slot] = topk_w[p];
    }
  }
  __syncthreads();
  // compact the active experts so the GEMV grids are P-sized (a full E-sized
  // grid costs millions of no-op block dispatches per step — measured 2.4×)
  if (threadIdx.x == 0) {
    int g = 0;
    for (int e = 0; e < E; ++e)
      if (counts[e] > 0) active[g++] = e;
    for (; g < T * K; ++g) active[g] = -1;
  }
}

// h[e*CAP + i][I] = silu(x_t · g_j) * (x_t · u_j); grid (E, I/4), 4 waves.
// CHUNKS = H / 512 (bf16x8 row-cache registers per lane); templated so the
// register file stays statically indexed (guide rule #20).
template <int CHUNKS>
__global__ __launch_bounds__(256)
void moe_gemv_h2_kernel(short* __restrict__ h,           // [E*CAP, I]
                        const short* __restrict__ x,      // [T, H]
                        const short* __restrict__ w13,    // [E, 2I, H]
                        const int* __restrict__ counts,
                        const int* __restrict__ tok_list,
                        const int* __restrict__ active,
                        int H, int I) {
  const int e = active[blockIdx.x];
  if (e < 0) return;
  const int n = counts[e];
  if (n == 0) return;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int j = blockIdx.y * 4 + wid;
  if (j >= I) return;
  const short* grow = w13 + ((long)e * 2 * I + j) * H;
  const short* urow = w13 + ((long)e * 2 * I + I + j) * H;

  // lane-interleaved layout: instruction c reads a contiguous 1 KiB
  // (64 lanes × 16 B) — lane-consecutive chunks were 4× uncoalesced
  bf16x8 greg[CHUNKS], ureg[CHUNKS];
  #pragma unroll
  for (int c = 0; c < CHUNKS; ++c) {
    greg[c] = *reinterpret_cast<const bf16x8*>(grow + (c * WAVE + lane) * 8);
    ureg[c] = *reinterpret_cast<const bf16x8*>(urow + (c * WAVE + lane) * 8);
  }
  const int lim = min(n, MD_CAP);
  for (int i = 0; i < lim; ++i) {
    const int t = tok_list[e * MD_CAP + i];
    const short* xrow = x + (long)t * H;
    float dg = 0.f, du = 0.f;
    #pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(xrow + (c * WAVE + lane) * 8);
      #pragma unroll
      for (int q_ = 0; q_ < 8; ++q_) {
        const float xf = bf2f(xv[q_]);
        dg += xf * bf2f(greg[c][q_]);
        du += xf * bf2f(ureg[c][q_]);
      }
    }
    dg = wave_reduce_sum(dg);
    du = wave_reduce_sum(du);
    if (lane == 0) {
      const float s = dg / (1.0f + __expf(-dg));
      h[((long)e * MD_CAP + i) * I + j] = f2bf(s * du);
    }
  }
}

// out[t] += w_p * (h_row · w2_row_o); grid (E, H/4), 4 waves.
__global__ __launch_bounds__(256)
void moe_gemv_down2_kernel(float* __restrict__ out,       // [T, H] f32
                           const short* __restrict__ h,    // [E*CAP, I]
                           const short* __restrict__ w2,   // [E, H, I]
                           const int* __restrict__ counts,
                           const int* __restrict__ tok_list,
                           const float* __restrict__ w_list,
                           const int* __restrict__ active,
                           int H, int I) {
  const int e = active[blockIdx.x];
  if (e < 0) return;
  const int n = counts[e];
  if (n == 0) return;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int o = blockIdx.y * 4 + wid;
  if (o >= H) return;
  const short* wrow = w2 + ((long)e * H + o) * I;
  // cache the down row: I/64 bf16 per lane (12 for I=768, 4 for I=256);
  // fully unrolled with a static guard so wreg stays in registers (rule #20)
  const int per_lane = I / WAVE;
  float wreg[16];
  #pragma unroll
  for (int c = 0; c < 16; ++c)   // lane-interleaved: coalesced per instruction
    wreg[c] = (c < per_lane) ? bf2f(wrow[c * WAVE + lane]) : 0.f;
  const int lim = min(n, MD_CAP);
  for (int i = 0; i < lim; ++i) {
    const short* hrow = h + ((long)e * MD_CAP + i) * I;
    float d = 0.f;
    #pragma unroll
    for (int c = 0; c < 16; ++c)
      if (c < per_lane) d += wreg[c] * bf2f(hrow[c * WAVE + lane]);
    d = wave_reduce_sum(d);
    if (lane == 0) {
      const int t = tok_list[e * MD_CAP + i];
      atomicAdd(out + (long)t * H + o, d * w_list[e * MD_CAP + i]);
    }
  }
}

void moe_gemv_dedup(torch::Tensor out, torch::Tensor x, torch::Tensor w13,
                    torch::Tensor w2, torch::Tensor topk_ids,
                    torch::Tensor topk_w, torch::Tensor counts,
                    torch::Tensor tok_list, torch::Tensor w_list,
                    torch::Tensor h, torch::Tensor active) {
  const int T = x.size(0), H = x.size(1);
  const int K = topk_ids.size(1), E = w13.size(0);
  const int I = w2.size(2);
  TORCH_CHECK(T * K <= 128 * MD_CAP && out.dtype() == torch::kFloat32);
  TORCH_CHECK(tok_list.size(0) == E && tok_list.size(1) == MD_CAP);
  TORCH_CHECK(h.size(0) == (long)E * MD_CAP && h.size(1) == I);
  TORCH_CHECK(I % WAVE == 0 && I / WAVE <= 16 && H % (WAVE * 8) == 0);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  const int P = T * K;
  TORCH_CHECK(active.numel() >= P);
  hipLaunchKernelGGL(moe_build_lists_kernel, dim3(1), dim3(256), 0, s,
                     counts.data_ptr<int>(), tok_list.data_ptr<int>(),
                     w_list.data_ptr<float>(), active.data_ptr<int>(),
                     topk_ids.data_ptr<int>(), topk_w.data_ptr<float>(), T, K, E);
  HIP_CHECK_KERNEL();
  const int chunks = H / (WAVE * 8);
  TORCH_CHECK(chunks == 1 || chunks == 2 || chunks == 4,
              "moe_gemv_dedup supports H in {512, 1024, 2048}");
  #define LAUNCH_H2(C) hipLaunchKernelGGL(moe_gemv_h2_kernel<C>, \
      dim3(P, (I + 3) / 4), dim3(256), 0, s, (short*)h.data_ptr(), \
      (const short*)x.data_ptr(), (const short*)w13.data_ptr(), \
      counts.data_ptr<int>(), tok_list.data_ptr<int>(), \
      active.data_ptr<int>(), H, I)
  if (chunks == 1) LAUNCH_H2(1);
  else if (chunks == 2) LAUNCH_H2(2);
  else LAUNCH_H2(4);
  #undef LAUNCH_H2
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(moe_gemv_down2_kernel, dim3(P, (H + 3) / 4), dim3(256), 0, s,
                     out.data_ptr<float>(), (const short*)h.data_ptr(),
                     (const short*)w2.data_ptr(), counts.data_ptr<int>(),
                     tok_list.data_ptr<int>(), w_list.data_ptr<float>(),
                     active.data_ptr<int>(), H, I);
  HIP_CHECK_KERNEL();
}
