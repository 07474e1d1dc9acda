// HBM-resident semantic-memory vector store: batched cosine scores + top-k.
//
// Replaces the reference's sqlite-vec CPU scan (src/shared/embeddings.ts:16-27,
// db-queries.ts:995-1010 `vec_distance_cosine` over BLOB rows). The matrix
// lives in HBM as [N, 384] bf16, rows L2-normalized at insert time, so cosine
// = plain dot. BASELINE config 4 sizes this at 10M × 384 (7.4 GB bf16 — a
// fraction of 288 GB HBM3E).
//
// Kernel: wave-per-row dot (384 dims / 64 lanes = 6 bf16 = 3×bf16x2 loads per
// lane → coalesced 1536 B per wave-instruction step), grid-stride; per-block
// top-k into a global candidate buffer; final merge of (blocks × k)
// candidates by a single-block second kernel.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

#define VS_DIM 384
#define VS_MAXK 64
#define VS_BLOCK 256   // 4 waves

__global__ __launch_bounds__(VS_BLOCK)
void vs_score_topk_kernel(float* __restrict__ cand_v,   // [nblocks, K]
                          int* __restrict__ cand_i,     // [nblocks, K]
                          const short* __restrict__ mat,  // [N, 384] bf16 (L2-normed)
                          const float* __restrict__ query,  // [384] (L2-normed)
                          int N, int K) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int waves_per_grid = gridDim.x * (VS_BLOCK / WAVE);
  const int gwave = blockIdx.x * (VS_BLOCK / WAVE) + wid;

  // query → registers (6 f32 per lane)
  float qreg[6];
  #pragma unroll
  for (int j = 0; j < 6; ++j) qreg[j] = query[lane * 6 + j];

  // per-wave local top-K kept by lane0..K-1? Simpler: per-wave insertion into
  // LDS list guarded by lane 0 (scores arrive one per row iteration).
  __shared__ float lv[VS_BLOCK / WAVE][VS_MAXK];
  __shared__ int li[VS_BLOCK / WAVE][VS_MAXK];
  if (lane < K) { lv[wid][lane] = -INFINITY; li[wid][lane] = -1; }

  float lmin = -INFINITY;
  for (long r = gwave; r < N; r += waves_per_grid) {
    const short* mrow = mat + r * VS_DIM + lane * 6;
    float dot = 0.f;
    // 6 bf16 per lane: one 4-wide + one 2-wide load
    bf16x4 a = *reinterpret_cast<const bf16x4*>(mrow);
    short b0 = mrow[4], b1 = mrow[5];
    dot = qreg[0] * bf2f(a[0]) + qreg[1] * bf2f(a[1]) + qreg[2] * bf2f(a[2])
        + qreg[3] * bf2f(a[3]) + qreg[4] * bf2f(b0) + qreg[5] * bf2f(b1);
    dot = wave_reduce_sum(dot);
    if (lane == 0 && dot > lmin) {
      int pos = K - 1;
      while (pos > 0 && lv[wid][pos - 1] < dot) {
        lv[wid][pos] = lv[wid][pos - 1]; li[wid][pos] = li[wid][pos - 1]; --pos;
      }
      lv[wid][pos] = dot; li[wid][pos] = (int)r;
      lmin = lv[wid][K - 1];
    }
    lmin = __shfl(lmin, 0, WAVE);
  }
  __syncthreads();

  // merge the block's 4 wave-lists (sorted desc) → block top-K by wave 0
  if (wid == 0 && lane == 0) {
    int p[VS_BLOCK / WAVE] = {0, 0, 0, 0};
    for (int k = 0; k < K; ++k) {
      float best = -INFINITY; int bw = 0;
      #pragma unroll
      for (int w = 0; w < VS_BLOCK / WAVE; ++w) {
        if (p[w] < K && lv[w][p[w]] > best) { best = lv[w][p[w]]; bw = w; }
      }
      cand_v[(long)blockIdx.x * K + k] = best;
      cand_i[(long)blockIdx.x * K + k] = (best == -INFINITY) ? -1 : li[bw][p[bw]];
      if (best != -INFINITY) ++p[bw];
    }
  }
}

// final merge: one block, iterative argmax over nblocks*K candidates (small)
__global__ void vs_merge_kernel(float* __restrict__ out_v,  // [K]
                                long* __restrict__ out_i,   // [K]
                                float* __restrict__ cand_v,
                                int* __restrict__ cand_i,
                                int ncand, int K) {
  const int lane = threadIdx.x & 63;
  if (threadIdx.x >= WAVE) return;
  for (int k = 0; k < K; ++k) {
    float best = -INFINITY; int besti = -1;
    for (int i = lane; i < ncand; i += WAVE) {
      if (cand_v[i] > best) { best = cand_v[i]; besti = i; }
    }
    #pragma unroll
    for (int o = 32; o > 0; o >>= 1) {
      float ov = __shfl_xor(best, o, WAVE);
      int oi = __shfl_xor(besti, o, WAVE);
      if (ov > best) { best = ov; besti = oi; }
    }
    besti = __shfl(besti, 0, WAVE);   // consistent winner everywhere
    if (lane == 0) {
      out_v[k] = best;
      out_i[k] = (besti >= 0) ? (long)cand_i[besti] : -1;
    }
    if (lane == 0 && besti >= 0) cand_v[besti] = -INFINITY;
  }
}

void vs_topk(torch::Tensor out_v, torch::Tensor out_i, torch::Tensor cand_v,
             torch::Tensor cand_i, torch::Tensor mat, torch::Tensor query,
             int64_t K) {
  TORCH_CHECK(mat.dtype() == torch::kBFloat16 && mat.size(1) == VS_DIM);
  TORCH_CHECK(query.dtype() == torch::kFloat32 && query.numel() == VS_DIM);
  TORCH_CHECK(K >= 1 && K <= VS_MAXK);
  const long N = mat.size(0);
  const int nblocks = (int)cand_v.size(0);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(vs_score_topk_kernel, dim3(nblocks), dim3(VS_BLOCK), 0, s,
                     cand_v.data_ptr<float>(), cand_i.data_ptr<int>(),
                     (const short*)mat.data_ptr(), query.data_ptr<float>(),
                     (int)N, (int)K);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(vs_merge_kernel, dim3(1), dim3(WAVE), 0, s,
                     out_v.data_ptr<float>(), out_i.data_ptr<long>(),
                     cand_v.data_ptr<float>(), cand_i.data_ptr<int>(),
                     nblocks * (int)K, (int)K);
  HIP_CHECK_KERNEL();
}
