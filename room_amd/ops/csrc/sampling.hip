// Fused sampling: temperature + top-k + top-p + multinomial draw, one kernel.
//
// The reference's local-model path delegated sampling to Ollama; here it is a
// CDNA4 kernel so a decode step never leaves the GPU (no logits→host copy).
// Vocab ≈ 152k. Scheme:
//   1) each of 256 threads keeps a sorted local top-K over its strided slice
//      (insertion guarded by the current min → ~O(1) amortized per element);
//   2) tournament merge in LDS: 8 rounds of pairwise sorted-list merges
//      (keep top-K), leaving the global top-K sorted at list 0;
//   3) thread 0 applies temperature softmax, top-p cut, and draws.
// LDS: 256 lists × K(≤64) × 8 B = ≤128 KiB (fits the 160 KiB/CU budget; this
// kernel runs at low occupancy by design — it is launch-latency bound).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

#define SMP_THREADS 256
#define SMP_MAXK 64

__device__ __forceinline__ float xorshift_unit(uint64_t* state) {
  uint64_t x = *state;
  x ^= x << 13; x ^= x >> 7; x ^= x << 17;
  *state = x;
  return (float)((x >> 11) & 0xFFFFFF) / 16777216.0f;  // [0,1)
}

__global__ __launch_bounds__(SMP_THREADS, 1)
void sample_kernel(int* __restrict__ out_tokens,      // [B]
                   const float* __restrict__ logits,   // [B, V]
                   uint64_t* __restrict__ seeds,        // [B] in/out RNG state
                   int V, int K, float temperature, float top_p) {
  const int b = blockIdx.x;
  const float* row = logits + (long)b * V;
  const int tid = threadIdx.x;

  __shared__ float cv[SMP_THREADS * SMP_MAXK];
  __shared__ int ci[SMP_THREADS * SMP_MAXK];

  // 1) local sorted-descending top-K over strided slice
  float lv[SMP_MAXK];
  int li[SMP_MAXK];
  for (int i = 0; i < K; ++i) { lv[i] = -INFINITY; li[i] = -1; }
  float lmin = -INFINITY;  // register copy of lv[K-1] (lv lives in scratch)
  for (int v = tid; v < V; v += SMP_THREADS) {
    float x = row[v];
    if (x <= lmin) continue;
    int pos = K - 1;
    while (pos > 0 && lv[pos - 1] < x) {
      lv[pos] = lv[pos - 1]; li[pos] = li[pos - 1]; --pos;
    }
    lv[pos] = x; li[pos] = v;
    lmin = lv[K - 1];
  }
  float* mycv = cv + tid * SMP_MAXK;
  int* myci = ci + tid * SMP_MAXK;
  for (int i = 0; i < K; ++i) { mycv[i] = lv[i]; myci[i] = li[i]; }
  __syncthreads();

  // 2) tournament merge: 8 rounds; thread t < half merges list[t+half] into
  //    list[t], both sorted desc → sorted top-K kept in registers then stored.
  for (int half = SMP_THREADS / 2; half >= 1; half >>= 1) {
    if (tid < half) {
      const float* av = cv + tid * SMP_MAXK;
      const int* ai = ci + tid * SMP_MAXK;
      const float* bv = cv + (tid + half) * SMP_MAXK;
      const int* bi = ci + (tid + half) * SMP_MAXK;
      float mv[SMP_MAXK];
      int mi[SMP_MAXK];
      int pa = 0, pb = 0;
      for (int i = 0; i < K; ++i) {
        if (pb >= K || (pa < K && av[pa] >= bv[pb])) {
          mv[i] = av[pa]; mi[i] = ai[pa]; ++pa;
        } else {
          mv[i] = bv[pb]; mi[i] = bi[pb]; ++pb;
        }
      }
      float* ov = cv + tid * SMP_MAXK;
      int* oi = ci + tid * SMP_MAXK;
      for (int i = 0; i < K; ++i) { ov[i] = mv[i]; oi[i] = mi[i]; }
    }
    __syncthreads();
  }

  // 3) thread 0: temperature softmax over global top-K (sorted desc at list 0),
  //    top-p nucleus cut, multinomial draw.
  if (tid == 0) {
    int n = 0;
    while (n < K && ci[n] >= 0 && cv[n] != -INFINITY) ++n;
    if (n == 0) { out_tokens[b] = 0; return; }
    if (temperature <= 1e-5f) { out_tokens[b] = ci[0]; return; }  // greedy
    const float invt = 1.0f / temperature;
    const float m = cv[0];
    float probs[SMP_MAXK];
    float denom = 0.f;
    for (int i = 0; i < n; ++i) {
      probs[i] = __expf((cv[i] - m) * invt);
      denom += probs[i];
    }
    float cum = 0.f;
    int cut = n;
    for (int i = 0; i < n; ++i) {
      cum += probs[i] / denom;
      if (cum >= top_p) { cut = i + 1; break; }
    }
    float denom2 = 0.f;
    for (int i = 0; i < cut; ++i) denom2 += probs[i];
    // stateful on-device RNG: the state advances in place, so graph replays
    // (hipGraph decode capture) draw fresh randomness with zero host work
    uint64_t st = seeds[b] | 1ull;
    float r = xorshift_unit(&st) * denom2;
    seeds[b] = st;
    float acc = 0.f;
    int pick = ci[cut - 1];
    for (int i = 0; i < cut; ++i) {
      acc += probs[i];
      if (r <= acc) { pick = ci[i]; break; }
    }
    out_tokens[b] = pick;
  }
}

void sample_tokens(torch::Tensor out_tokens, torch::Tensor logits,
                   torch::Tensor seeds, int64_t top_k, double temperature,
                   double top_p) {
  const int B = logits.size(0), V = logits.size(1);
  TORCH_CHECK(logits.dtype() == torch::kFloat32);
  TORCH_CHECK(top_k >= 1 && top_k <= SMP_MAXK);
  dim3 grid(B), block(SMP_THREADS);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(sample_kernel, grid, block, 0, s,
                     out_tokens.data_ptr<int>(), logits.data_ptr<float>(),
                     (uint64_t*)seeds.data_ptr(), V, (int)top_k,
                     (float)temperature, (float)top_p);
  HIP_CHECK_KERNEL();
}
