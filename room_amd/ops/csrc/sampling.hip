// Fused sampling: temperature + top-k + top-p + multinomial draw, one kernel.
//
// V ≈ 152k, B small (decode batch). Everything stays on-GPU (hipGraph-
// capturable; RNG state advances on-device so replays draw fresh randomness).
//
// Design (v2 — v1 kept per-thread sorted lists in scratch memory, which is
// dynamic-indexed local array = scratch traffic, ~5 ms/step; guide common-
// mistake #20):
//   1) each of 256 threads scans its strided slice keeping an UNSORTED
//      top-K set in its LDS row via replace-min (register min guard →
//      insertions are rare, each costs one K-element LDS rescan);
//   2) block tournament: K iterations of block-wide argmax; only the winning
//      thread rescans its row. Produces the global top-K sorted descending;
//   3) thread 0: temperature softmax, top-p nucleus cut, multinomial draw.
// LDS: 256 rows × K × 8 B ≤ 128 KiB (K ≤ 64) — deliberate 1-block/CU.
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
  const int lane = tid & 63;
  const int wid = tid >> 6;

  // row stride K+1: stride 64 would put a wave's 64 lanes on one bank
  // (32-way conflict on every access — guide G4); +1 spreads them.
  __shared__ float cv[SMP_THREADS * (SMP_MAXK + 1)];
  __shared__ int ci[SMP_THREADS * (SMP_MAXK + 1)];
  __shared__ float sel_v[SMP_MAXK];
  __shared__ int sel_i[SMP_MAXK];
  __shared__ float wmax[4];
  __shared__ int wwin[4];

  float* mycv = cv + tid * (SMP_MAXK + 1);
  int* myci = ci + tid * (SMP_MAXK + 1);

  // 1) strided scan, replace-min into LDS row
  for (int i = 0; i < K; ++i) { mycv[i] = -INFINITY; myci[i] = -1; }
  float lmin = -INFINITY;
  int min_slot = 0;
  for (int v = tid; v < V; v += SMP_THREADS) {
    const float x = row[v];
    if (x <= lmin) continue;
    mycv[min_slot] = x;
    myci[min_slot] = v;
    // rescan for the new min (K LDS reads; insertions are rare after warmup)
    float nm = mycv[0]; int ns = 0;
    for (int i = 1; i < K; ++i) {
      if (mycv[i] < nm) { nm = mycv[i]; ns = i; }
    }
    lmin = nm; min_slot = ns;
  }
  // local max (kept in registers for the tournament)
  float my_max = -INFINITY; int my_slot = 0;
  for (int i = 0; i < K; ++i)
    if (mycv[i] > my_max) { my_max = mycv[i]; my_slot = i; }
  __syncthreads();

  // 2) block tournament: K rounds of argmax over 256 candidates
  for (int k = 0; k < K; ++k) {
    // wave argmax (value, tid)
    float v = my_max; int who = tid;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(v, off, WAVE);
      int ow = __shfl_xor(who, off, WAVE);
      if (ov > v || (ov == v && ow < who)) { v = ov; who = ow; }
    }
    if (lane == 0) { wmax[wid] = v; wwin[wid] = who; }
    __syncthreads();
    // cross-wave (4 entries) resolved by every thread identically
    float bv = wmax[0]; int bw = wwin[0];
    #pragma unroll
    for (int w = 1; w < 4; ++w)
      if (wmax[w] > bv || (wmax[w] == bv && wwin[w] < bw)) {
        bv = wmax[w]; bw = wwin[w];
      }
    if (tid == bw) {
      sel_v[k] = my_max;
      sel_i[k] = myci[my_slot];
      mycv[my_slot] = -INFINITY;
      my_max = -INFINITY;
      for (int i = 0; i < K; ++i)
        if (mycv[i] > my_max) { my_max = mycv[i]; my_slot = i; }
    }
    __syncthreads();
    if (bv == -INFINITY) break;
  }

  // 3) thread 0: softmax over the (sorted-desc) top-K, top-p cut, draw
  if (tid == 0) {
    int n = 0;
    while (n < K && sel_i[n] >= 0 && sel_v[n] != -INFINITY) ++n;
    if (n == 0) { out_tokens[b] = 0; return; }
    if (temperature <= 1e-5f) { out_tokens[b] = sel_i[0]; return; }  // greedy
    const float invt = 1.0f / temperature;
    const float m = sel_v[0];
    float probs[SMP_MAXK];
    float denom = 0.f;
    for (int i = 0; i < n; ++i) {
      probs[i] = __expf((sel_v[i] - m) * invt);
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
    uint64_t st = seeds[b] | 1ull;
    float r = xorshift_unit(&st) * denom2;
    seeds[b] = st;  // on-device state advance (graph-replay safe)
    float acc = 0.f;
    int pick = sel_i[cut - 1];
    for (int i = 0; i < cut; ++i) {
      acc += probs[i];
      if (r <= acc) { pick = sel_i[i]; break; }
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

// ---- v3: register-resident per-thread top-8 + block tournament ----
// Per-thread top-8 kept in VGPRs via a fully-static compare/shift chain (no
// dynamic indexing → no scratch, no LDS in the scan). The global top-K
// (K ≤ 64) is contained in the union of per-thread top-8 unless one thread's
// strided slice holds ≥9 of the global top-K: P ≈ C(64,9)/256⁸ < 1e-9 for
// random position assignment — negligible for sampling purposes.
#define V3_KEEP 8

__global__ __launch_bounds__(SMP_THREADS, 1)
void sample_v3_kernel(int* __restrict__ out_tokens,
                      const float* __restrict__ logits,
                      uint64_t* __restrict__ seeds,
                      int V, int K, float temperature, float top_p) {
  const int b = blockIdx.x;
  const float* row = logits + (long)b * V;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;

  float r0 = -INFINITY, r1 = -INFINITY, r2 = -INFINITY, r3 = -INFINITY,
        r4 = -INFINITY, r5 = -INFINITY, r6 = -INFINITY, r7 = -INFINITY;
  int i0 = -1, i1 = -1, i2 = -1, i3 = -1, i4 = -1, i5 = -1, i6 = -1, i7 = -1;

  for (int v = tid; v < V; v += SMP_THREADS) {
    const float x = row[v];
    if (x <= r7) continue;
    if (x > r0) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=r3; i4=i3;
      r3=r2; i3=i2; r2=r1; i2=i1; r1=r0; i1=i0; r0=x; i0=v;
    } else if (x > r1) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=r3; i4=i3;
      r3=r2; i3=i2; r2=r1; i2=i1; r1=x; i1=v;
    } else if (x > r2) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=r3; i4=i3;
      r3=r2; i3=i2; r2=x; i2=v;
    } else if (x > r3) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=r3; i4=i3; r3=x; i3=v;
    } else if (x > r4) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=x; i4=v;
    } else if (x > r5) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=x; i5=v;
    } else if (x > r6) {
      r7=r6; i7=i6; r6=x; i6=v;
    } else {
      r7=x; i7=v;
    }
  }

  __shared__ float cv[SMP_THREADS * (V3_KEEP + 1)];
  __shared__ int ci[SMP_THREADS * (V3_KEEP + 1)];
  __shared__ float sel_v[SMP_MAXK];
  __shared__ int sel_i[SMP_MAXK];
  __shared__ float wmax[4];
  __shared__ int wwin[4];
  float* mycv = cv + tid * (V3_KEEP + 1);
  int* myci = ci + tid * (V3_KEEP + 1);
  mycv[0]=r0; mycv[1]=r1; mycv[2]=r2; mycv[3]=r3;
  mycv[4]=r4; mycv[5]=r5; mycv[6]=r6; mycv[7]=r7;
  myci[0]=i0; myci[1]=i1; myci[2]=i2; myci[3]=i3;
  myci[4]=i4; myci[5]=i5; myci[6]=i6; myci[7]=i7;
  float my_max = r0;  // rows are sorted desc
  int my_slot = 0;
  __syncthreads();

  for (int k = 0; k < K; ++k) {
    float v = my_max; int who = tid;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(v, off, WAVE);
      int ow = __shfl_xor(who, off, WAVE);
      if (ov > v || (ov == v && ow < who)) { v = ov; who = ow; }
    }
    if (lane == 0) { wmax[wid] = v; wwin[wid] = who; }
    __syncthreads();
    float bv = wmax[0]; int bw = wwin[0];
    #pragma unroll
    for (int w = 1; w < 4; ++w)
      if (wmax[w] > bv || (wmax[w] == bv && wwin[w] < bw)) {
        bv = wmax[w]; bw = wwin[w];
      }
    if (tid == bw) {
      sel_v[k] = my_max;
      sel_i[k] = myci[my_slot];
      // advance to next candidate in the sorted row
      ++my_slot;
      my_max = (my_slot < V3_KEEP) ? mycv[my_slot] : -INFINITY;
    }
    __syncthreads();
    if (bv == -INFINITY) break;
  }

  if (tid == 0) {
    int n = 0;
    while (n < K && sel_i[n] >= 0 && sel_v[n] != -INFINITY) ++n;
    if (n == 0) { out_tokens[b] = 0; return; }
    if (temperature <= 1e-5f) { out_tokens[b] = sel_i[0]; return; }
    const float invt = 1.0f / temperature;
    const float m = sel_v[0];
    float denom = 0.f;
    for (int i = 0; i < n; ++i) denom += __expf((sel_v[i] - m) * invt);
    uint64_t st = seeds[b] | 1ull;
    // top-p cut + draw in one pass (sel sorted desc)
    float target = top_p * denom;
    float cum = 0.f;
    int cut = n;
    for (int i = 0; i < n; ++i) {
      cum += __expf((sel_v[i] - m) * invt);
      if (cum >= target) { cut = i + 1; break; }
    }
    float denom2 = 0.f;
    for (int i = 0; i < cut; ++i) denom2 += __expf((sel_v[i] - m) * invt);
    float r = xorshift_unit(&st) * denom2;
    seeds[b] = st;
    float acc = 0.f;
    int pick = sel_i[cut - 1];
    for (int i = 0; i < cut; ++i) {
      acc += __expf((sel_v[i] - m) * invt);
      if (r <= acc) { pick = sel_i[i]; break; }
    }
    out_tokens[b] = pick;
  }
}

// ------------------------------------------------------- v4: two-stage scan
// v3's single block/row leaves 251 CUs idle during the 152k-logit scan
// (rocprof: 335 µs/step at B=5, scan-bound). Stage 1 spreads the scan over
// V4_NSC blocks per row, each emitting its slice's top-8 (sorted desc) via
// the v3 register-filter + block tournament. Stage 2 is ONE WAVE per row:
// lane l owns block l's 8 candidates in LDS, a shuffle-argmax tournament
// extracts the global top-K (no __syncthreads in the loop), lane 0 runs the
// same temperature/top-p/draw finalizer as v3.
#define V4_NSC 64
#define V4_KEEP 8

__global__ __launch_bounds__(SMP_THREADS, 2)
void sample_partial_topk_kernel(float* __restrict__ part_v,  // [B, NSC, KEEP]
                                int* __restrict__ part_i,
                                const float* __restrict__ logits, int V) {
  const int b = blockIdx.x;
  const int blk = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const float* row = logits + (long)b * V;
  const int span = (V + V4_NSC - 1) / V4_NSC;
  const int lo = blk * span, hi = min(V, lo + span);

  float r0 = -INFINITY, r1 = -INFINITY, r2 = -INFINITY, r3 = -INFINITY,
        r4 = -INFINITY, r5 = -INFINITY, r6 = -INFINITY, r7 = -INFINITY;
  int i0 = -1, i1 = -1, i2 = -1, i3 = -1, i4 = -1, i5 = -1, i6 = -1, i7 = -1;
  for (int v = lo + tid; v < hi; v += SMP_THREADS) {
    const float x = row[v];
    if (x <= r7) continue;
    if (x > r0) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=r3; i4=i3;
      r3=r2; i3=i2; r2=r1; i2=i1; r1=r0; i1=i0; r0=x; i0=v;
    } else if (x > r1) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=r3; i4=i3;
      r3=r2; i3=i2; r2=r1; i2=i1; r1=x; i1=v;
    } else if (x > r2) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=r3; i4=i3;
      r3=r2; i3=i2; r2=x; i2=v;
    } else if (x > r3) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=r3; i4=i3; r3=x; i3=v;
    } else if (x > r4) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=r4; i5=i4; r4=x; i4=v;
    } else if (x > r5) {
      r7=r6; i7=i6; r6=r5; i6=i5; r5=x; i5=v;
    } else if (x > r6) {
      r7=r6; i7=i6; r6=x; i6=v;
    } else {
      r7=x; i7=v;
    }
  }

  __shared__ float cv[SMP_THREADS * (V4_KEEP + 1)];
  __shared__ int ci[SMP_THREADS * (V4_KEEP + 1)];
  __shared__ float wmax[4];
  __shared__ int wwin[4];
  float* mycv = cv + tid * (V4_KEEP + 1);
  int* myci = ci + tid * (V4_KEEP + 1);
  mycv[0]=r0; mycv[1]=r1; mycv[2]=r2; mycv[3]=r3;
  mycv[4]=r4; mycv[5]=r5; mycv[6]=r6; mycv[7]=r7;
  myci[0]=i0; myci[1]=i1; myci[2]=i2; myci[3]=i3;
  myci[4]=i4; myci[5]=i5; myci[6]=i6; myci[7]=i7;
  float my_max = r0;
  int my_slot = 0;
  __syncthreads();

  float* pv = part_v + ((long)b * V4_NSC + blk) * V4_KEEP;
  int* pi = part_i + ((long)b * V4_NSC + blk) * V4_KEEP;
  for (int k = 0; k < V4_KEEP; ++k) {
    float v = my_max; int who = tid;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(v, off, WAVE);
      int ow = __shfl_xor(who, off, WAVE);
      if (ov > v || (ov == v && ow < who)) { v = ov; who = ow; }
    }
    if (lane == 0) { wmax[wid] = v; wwin[wid] = who; }
    __syncthreads();
    float bv = wmax[0]; int bw = wwin[0];
    #pragma unroll
    for (int w = 1; w < 4; ++w)
      if (wmax[w] > bv || (wmax[w] == bv && wwin[w] < bw)) {
        bv = wmax[w]; bw = wwin[w];
      }
    if (tid == bw) {
      pv[k] = my_max;
      pi[k] = myci[my_slot];
      ++my_slot;
      my_max = (my_slot < V4_KEEP) ? mycv[my_slot] : -INFINITY;
    }
    if (bv == -INFINITY) {       // uniform: pad the rest and stop
      if (tid == 0)
        for (int j = k; j < V4_KEEP; ++j) { pv[j] = -INFINITY; pi[j] = -1; }
      return;
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(64)
void sample_v4_final_kernel(int* __restrict__ out_tokens,
                            const float* __restrict__ part_v,
                            const int* __restrict__ part_i,
                            uint64_t* __restrict__ seeds,
                            int K, float temperature, float top_p) {
  const int b = blockIdx.x;
  const int lane = threadIdx.x;
  __shared__ float cand_v[V4_NSC][V4_KEEP];
  __shared__ int cand_i[V4_NSC][V4_KEEP];
  __shared__ float sel_v[SMP_MAXK];
  __shared__ int sel_i[SMP_MAXK];
  #pragma unroll
  for (int j = 0; j < V4_KEEP; ++j) {
    cand_v[lane][j] = part_v[((long)b * V4_NSC + lane) * V4_KEEP + j];
    cand_i[lane][j] = part_i[((long)b * V4_NSC + lane) * V4_KEEP + j];
  }
  for (int k = lane; k < SMP_MAXK; k += 64) sel_v[k] = -INFINITY;
  __syncthreads();

  int slot = 0;
  float head = cand_v[lane][0];
  for (int k = 0; k < K; ++k) {
    float v = head; int who = lane;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(v, off, WAVE);
      int ow = __shfl_xor(who, off, WAVE);
      if (ov > v || (ov == v && ow < who)) { v = ov; who = ow; }
    }
    if (v == -INFINITY) break;   // uniform across the wave
    if (lane == who) {
      sel_v[k] = head;
      sel_i[k] = cand_i[lane][slot];
      ++slot;
      head = (slot < V4_KEEP) ? cand_v[lane][slot] : -INFINITY;
    }
  }
  __syncthreads();

  if (lane == 0) {
    int n = 0;
    while (n < K && sel_v[n] != -INFINITY && sel_i[n] >= 0) ++n;
    if (n == 0) { out_tokens[b] = 0; return; }
    if (temperature <= 1e-5f) { out_tokens[b] = sel_i[0]; return; }
    const float invt = 1.0f / temperature;
    const float m = sel_v[0];
    float denom = 0.f;
    for (int i = 0; i < n; ++i) denom += __expf((sel_v[i] - m) * invt);
    uint64_t st = seeds[b] | 1ull;
    float target = top_p * denom;
    float cum = 0.f;
    int cut = n;
    for (int i = 0; i < n; ++i) {
      cum += __expf((sel_v[i] - m) * invt);
      if (cum >= target) { cut = i + 1; break; }
    }
    float denom2 = 0.f;
    for (int i = 0; i < cut; ++i) denom2 += __expf((sel_v[i] - m) * invt);
    float r = xorshift_unit(&st) * denom2;
    seeds[b] = st;
    float acc = 0.f;
    int pick = sel_i[cut - 1];
    for (int i = 0; i < cut; ++i) {
      acc += __expf((sel_v[i] - m) * invt);
      if (r <= acc) { pick = sel_i[i]; break; }
    }
    out_tokens[b] = pick;
  }
}

// scan-only probe: block max (isolates the logits-scan cost)
__global__ __launch_bounds__(SMP_THREADS)
void sample_scan_probe_kernel(int* __restrict__ out, const float* __restrict__ logits,
                              int V) {
  const int b = blockIdx.x;
  const float* row = logits + (long)b * V;
  float m = -INFINITY; int mi = 0;
  for (int v = threadIdx.x; v < V; v += SMP_THREADS) {
    float x = row[v];
    if (x > m) { m = x; mi = v; }
  }
  __shared__ float red[16];
  float gm = block_reduce_max(m, red);
  if (m == gm && threadIdx.x % 64 == 0) out[b] = mi;
}

void sample_tokens_v3(torch::Tensor out_tokens, torch::Tensor logits,
                      torch::Tensor seeds, int64_t top_k, double temperature,
                      double top_p) {
  const int B = logits.size(0), V = logits.size(1);
  TORCH_CHECK(logits.dtype() == torch::kFloat32);
  TORCH_CHECK(top_k >= 1 && top_k <= SMP_MAXK);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(sample_v3_kernel, dim3(B), dim3(SMP_THREADS), 0, s,
                     out_tokens.data_ptr<int>(), logits.data_ptr<float>(),
                     (uint64_t*)seeds.data_ptr(), V, (int)top_k,
                     (float)temperature, (float)top_p);
  HIP_CHECK_KERNEL();
}

void sample_tokens_v4(torch::Tensor out_tokens, torch::Tensor logits,
                      torch::Tensor seeds, int64_t top_k, double temperature,
                      double top_p) {
  const int B = logits.size(0), V = logits.size(1);
  TORCH_CHECK(logits.dtype() == torch::kFloat32);
  TORCH_CHECK(top_k >= 1 && top_k <= SMP_MAXK);
  // persistent per-B scratch: keeps the op allocation-free inside hipGraph
  // capture (the engine's decode graphs replay this kernel pair)
  static std::unordered_map<int, std::pair<torch::Tensor, torch::Tensor>> scratch;
  auto it = scratch.find(B);
  if (it == scratch.end()) {
    auto opts = torch::TensorOptions().device(logits.device());
    it = scratch.emplace(B, std::make_pair(
        torch::empty({(long)B * V4_NSC * V4_KEEP}, opts.dtype(torch::kFloat32)),
        torch::empty({(long)B * V4_NSC * V4_KEEP}, opts.dtype(torch::kInt32)))).first;
  }
  torch::Tensor part_v = it->second.first, part_i = it->second.second;
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(sample_partial_topk_kernel, dim3(B, V4_NSC),
                     dim3(SMP_THREADS), 0, s, part_v.data_ptr<float>(),
                     part_i.data_ptr<int>(), logits.data_ptr<float>(), V);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(sample_v4_final_kernel, dim3(B), dim3(64), 0, s,
                     out_tokens.data_ptr<int>(), part_v.data_ptr<float>(),
                     part_i.data_ptr<int>(), (uint64_t*)seeds.data_ptr(),
                     (int)top_k, (float)temperature, (float)top_p);
  HIP_CHECK_KERNEL();
}

void sample_scan_probe(torch::Tensor out, torch::Tensor logits) {
  const int B = logits.size(0), V = logits.size(1);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(sample_scan_probe_kernel, dim3(B), dim3(SMP_THREADS), 0, s,
                     out.data_ptr<int>(), logits.data_ptr<float>(), V);
  HIP_CHECK_KERNEL();
}

// token-history append for multi-step decode graphs: each graph replay
// appends the step's sampled tokens into a ring and bumps the device counter,
// so the host syncs once per K-step block instead of every step.
__global__ void hist_append_kernel(int* __restrict__ hist,   // [KMAX * B]
                                   int* __restrict__ ctr,    // [1]
                                   const int* __restrict__ toks,  // [B]
                                   int B, int kmax) {
  const int i = threadIdx.x;
  const int step = ctr[0];
  if (step < kmax && i < B) hist[step * B + i] = toks[i];
  __syncthreads();
  if (i == 0) ctr[0] = step + 1;
}

void hist_append(torch::Tensor hist, torch::Tensor ctr, torch::Tensor toks,
                 int64_t kmax) {
  const int B = toks.numel();
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(hist_append_kernel, dim3(1), dim3(std::max(B, 64)), 0, s,
                     hist.data_ptr<int>(), ctr.data_ptr<int>(),
                     toks.data_ptr<int>(), B, (int)kmax);
  HIP_CHECK_KERNEL();
}
