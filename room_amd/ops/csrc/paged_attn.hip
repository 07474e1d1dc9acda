// Paged attention for GQA (Qwen3-30B-A3B: 32 q heads / 4 kv heads / head_dim 128).
//
// One unified kernel serves prefill AND decode: every query token attends to
// the paged KV cache of its sequence up to its own position (causal). Decode
// is the T == batch case. This keeps the agent-swarm engine simple: sessions
// are KV-cache residency (SURVEY §5 "session-as-KV-cache") and both phases
// read the same cache.
//
// Layouts (chosen for this kernel, we own the cache):
//   K,V cache: [num_blocks, kv_heads, BLOCK_SIZE, head_dim] bf16
//   block_table: [num_seqs, max_blocks] int32
//
// Workgroup = (query token, kv head): 256 threads = 4 waves.
//   score phase: thread (h = t>>5, j = t&31) computes dot(q[h], k[pos_j])
//                (full 128-dim dot per thread; rows stream through L1)
//   value phase: thread (h = t>>5, d4 = (t&31)*4) accumulates 4 output dims;
//                V reads are 32×8B consecutive = coalesced per head-group.
// Online softmax with per-head running max/sum (flash-style, fp32).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

#define BLOCK_SIZE 16
#define CHUNK 32          // kv positions per iteration
#define QH_PER_KV 8       // GQA group width (32/4)
#define HEAD_DIM 128

__global__ __launch_bounds__(256)
void paged_attn_kernel(short* __restrict__ out,          // [T, Hq, D]
                       const short* __restrict__ q,      // [T, Hq, D]
                       const short* __restrict__ kcache, // [NB, Hk, BS, D]
                       const short* __restrict__ vcache,
                       const int* __restrict__ block_table,  // [S, MB]
                       const int* __restrict__ seq_ids,      // [T]
                       const int* __restrict__ q_pos,        // [T] absolute pos
                       int n_kvheads, int max_blocks, int q_row_stride,
                       float scale) {
  const int t = blockIdx.x;       // query token
  const int hk = blockIdx.y;      // kv head
  const int tid = threadIdx.x;
  const int h = tid >> 5;         // 0..7: q-head within group
  const int sub = tid & 31;       // score: position lane / value: dim quarter
  const int seq = seq_ids[t];
  const int bound = q_pos[t] + 1; // causal: attend to [0, bound)

  const int hq = hk * QH_PER_KV + h;
  const int n_qheads = n_kvheads * QH_PER_KV;

  __shared__ float q_s[QH_PER_KV][HEAD_DIM];
  __shared__ float p_s[QH_PER_KV][CHUNK];

  // load q for the 8 heads of this group into LDS (fp32)
  for (int i = tid; i < QH_PER_KV * HEAD_DIM; i += blockDim.x) {
    int hh = i / HEAD_DIM, dd = i % HEAD_DIM;
    q_s[hh][dd] = bf2f(q[(long)t * q_row_stride
                         + (hk * QH_PER_KV + hh) * HEAD_DIM + dd]);
  }
  __syncthreads();

  float acc[4] = {0.f, 0.f, 0.f, 0.f};
  float m_run = -INFINITY, l_run = 0.f;

  const long kv_stride_block = (long)n_kvheads * BLOCK_SIZE * HEAD_DIM;
  const int* btab = block_table + (long)seq * max_blocks;

  for (int base = 0; base < bound; base += CHUNK) {
    // ---- score phase: this thread scores position base+sub for head h
    {
      const int pos = base + sub;
      float s = -INFINITY;
      if (pos < bound) {
        const int blk = btab[pos / BLOCK_SIZE];
        const short* krow = kcache + (long)blk * kv_stride_block
                            + ((long)hk * BLOCK_SIZE + (pos % BLOCK_SIZE)) * HEAD_DIM;
        float dot = 0.f;
        #pragma unroll
        for (int v8 = 0; v8 < HEAD_DIM / 8; ++v8) {
          bf16x8 kv = *reinterpret_cast<const bf16x8*>(krow + v8 * 8);
          #pragma unroll
          for (int j = 0; j < 8; ++j) dot += q_s[h][v8 * 8 + j] * bf2f(kv[j]);
        }
        s = dot * scale;
      }
      p_s[h][sub] = s;
    }
    __syncthreads();

    // ---- softmax + value phase: thread owns dims [sub*4, sub*4+4) of head h
    {
      float chunk_max = -INFINITY;
      #pragma unroll
      for (int j = 0; j < CHUNK; ++j) chunk_max = fmaxf(chunk_max, p_s[h][j]);
      const float m_new = fmaxf(m_run, chunk_max);
      if (m_new != -INFINITY) {
        const float rescale = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
        acc[0] *= rescale; acc[1] *= rescale; acc[2] *= rescale; acc[3] *= rescale;
        l_run *= rescale;
        const int lim = min(CHUNK, bound - base);
        for (int j = 0; j < lim; ++j) {
          const float w = __expf(p_s[h][j] - m_new);
          l_run += w;
          const int pos = base + j;
          const int blk = btab[pos / BLOCK_SIZE];
          const short* vrow = vcache + (long)blk * kv_stride_block
                              + ((long)hk * BLOCK_SIZE + (pos % BLOCK_SIZE)) * HEAD_DIM
                              + sub * 4;
          bf16x4 vv = *reinterpret_cast<const bf16x4*>(vrow);
          acc[0] += w * bf2f(vv[0]);
          acc[1] += w * bf2f(vv[1]);
          acc[2] += w * bf2f(vv[2]);
          acc[3] += w * bf2f(vv[3]);
        }
        m_run = m_new;
      }
    }
    __syncthreads();
  }

  const float inv_l = (l_run > 0.f) ? 1.0f / l_run : 0.f;
  short* orow = out + ((long)t * n_qheads + hq) * HEAD_DIM + sub * 4;
  bf16x4 o;
  #pragma unroll
  for (int j = 0; j < 4; ++j) o[j] = f2bf(acc[j] * inv_l);
  *reinterpret_cast<bf16x4*>(orow) = o;

  // NOTE: the value phase recomputes l_run identically in all 32 threads of a
  // head (each walks the same 32 scores) — redundant VALU, zero extra HBM.
}

// ------------------------------------------------ KV cache scatter
// Write new K/V rows for T tokens into their paged slots.
__global__ void write_kv_kernel(short* __restrict__ kcache,
                                short* __restrict__ vcache,
                                const short* __restrict__ k,   // [T, Hk, D]
                                const short* __restrict__ v,
                                const int* __restrict__ block_table,
                                const int* __restrict__ seq_ids,
                                const int* __restrict__ q_pos,
                                int n_kvheads, int max_blocks) {
  const int t = blockIdx.x;
  const int seq = seq_ids[t];
  const int pos = q_pos[t];
  const int blk = block_table[(long)seq * max_blocks + pos / BLOCK_SIZE];
  const long dst_base = ((long)blk * n_kvheads) * BLOCK_SIZE * HEAD_DIM
                        + (long)(pos % BLOCK_SIZE) * HEAD_DIM;
  // threads cover Hk * D elements in 8-wide vectors
  const int total_vec = n_kvheads * HEAD_DIM / 8;
  for (int i = threadIdx.x; i < total_vec; i += blockDim.x) {
    const int hh = (i * 8) / HEAD_DIM;
    const int dd = (i * 8) % HEAD_DIM;
    const long dst = dst_base + (long)hh * BLOCK_SIZE * HEAD_DIM + dd;
    *reinterpret_cast<bf16x8*>(kcache + dst) =
        *reinterpret_cast<const bf16x8*>(k + ((long)t * n_kvheads + hh) * HEAD_DIM + dd);
    *reinterpret_cast<bf16x8*>(vcache + dst) =
        *reinterpret_cast<const bf16x8*>(v + ((long)t * n_kvheads + hh) * HEAD_DIM + dd);
  }
}

// ROOMAMD_ATTN_CHUNK=64 selects the larger-chunk split kernel (fewer
// barriers per span, more LDS); read once, process-stable so hipGraph
// replay stays consistent.
static int attn_chunk() {
  static int c = [] {
    const char* e = getenv("ROOMAMD_ATTN_CHUNK");
    return (e && atoi(e) == 64) ? 64 : 32;
  }();
  return c;
}

static bool attn_pipe() {
  // measured SLOWER in-bench (decode 10.70 -> 10.95 s over 6 steps): the
  // staging registers + duplicated LDS buffers cost more occupancy than the
  // hidden HBM latency buys. Off by default; ROOMAMD_ATTN_PIPE=1 re-enables.
  static bool p = [] {
    const char* e = getenv("ROOMAMD_ATTN_PIPE");
    return e && e[0] == '1';
  }();
  return p;
}

// ---------------------------------------------------------------- wrappers

void paged_attention(torch::Tensor out, torch::Tensor q, torch::Tensor kcache,
                     torch::Tensor vcache, torch::Tensor block_table,
                     torch::Tensor seq_ids, torch::Tensor q_pos, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(q.size(-1) == HEAD_DIM, "head_dim must be 128");
  const int T = q.size(0);
  const int n_kvheads = kcache.size(1);
  TORCH_CHECK(q.size(1) == n_kvheads * QH_PER_KV, "GQA group must be 8");
  TORCH_CHECK(kcache.size(2) == BLOCK_SIZE);
  const int max_blocks = block_table.size(1);
  dim3 grid(T, n_kvheads), block(256);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(paged_attn_kernel, grid, block, 0, s,
                     (short*)out.data_ptr(), (const short*)q.data_ptr(),
                     (const short*)kcache.data_ptr(), (const short*)vcache.data_ptr(),
                     block_table.data_ptr<int>(), seq_ids.data_ptr<int>(),
                     q_pos.data_ptr<int>(), n_kvheads, max_blocks,
                     (int)q.stride(0), (float)scale);
  HIP_CHECK_KERNEL();
}

void write_kv(torch::Tensor kcache, torch::Tensor vcache, torch::Tensor k,
              torch::Tensor v, torch::Tensor block_table, torch::Tensor seq_ids,
              torch::Tensor q_pos) {
  const int T = k.size(0);
  const int n_kvheads = kcache.size(1);
  const int max_blocks = block_table.size(1);
  dim3 grid(T), block(128);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(write_kv_kernel, grid, block, 0, s,
                     (short*)kcache.data_ptr(), (short*)vcache.data_ptr(),
                     (const short*)k.data_ptr(), (const short*)v.data_ptr(),
                     block_table.data_ptr<int>(), seq_ids.data_ptr<int>(),
                     q_pos.data_ptr<int>(), n_kvheads, max_blocks);
  HIP_CHECK_KERNEL();
}

// ------------------------------------------------ split-KV flash-decode
// Decode grids are tiny (T = batch ≈ swarm width, 4 kv heads → ~20 workgroups
// for 256 CUs). Split the KV range across NSPLITS workgroups per (token, kv
// head); each writes an unnormalized partial (acc, m, l); a merge kernel
// combines. NSPLITS is static so the decode step stays hipGraph-capturable.

// NSPLITS=64 A/B: wins only at L>=16k (63.6->50.0 us B=1), loses at the
// bench's 4-6k sessions (8.1->8.3 ms/step) and at B=1 (4.5->5.0): empty-split
// dispatch + 2x merge traffic. Static 32 keeps the decode graph replayable.
#define NSPLITS 32          // default split count (short contexts)
#define NSPLITS_MAX 64      // part-buffer layout stride; grid.z may be 32 or 64

// CH: kv positions staged/scored per iteration. 64 halves the barrier count
// per span (3 barriers per chunk) at the cost of LDS (41 KB vs 24 KB).
// PIPE: double-buffered K/V staging — the next chunk's global loads are
// issued right after the current chunk's LDS stores become visible, so the
// HBM latency overlaps the score+value phases instead of stalling at the
// top of each iteration (the round-1 in-phase interleave attempt failed
// because it stretched the phases' critical paths; this version keeps the
// phases untouched and only moves the load-issue point).
template <int CH, bool PIPE>
__global__ __launch_bounds__(256)
void paged_attn_split_kernel(float* __restrict__ part,   // [T, Hq, NSPLITS, D]
                             float* __restrict__ part_ml, // [T, Hq, NSPLITS, 2]
                             const short* __restrict__ q,
                             const short* __restrict__ kcache,
                             const short* __restrict__ vcache,
                             const int* __restrict__ block_table,
                             const int* __restrict__ seq_ids,
                             const int* __restrict__ q_pos,
                             int n_kvheads, int max_blocks, int q_row_stride,
                             float scale,
                             float* __restrict__ o_zero, long zero_n) {
  const int t = blockIdx.x;
  const int hk = blockIdx.y;
  const int split = blockIdx.z;
  const int tid = threadIdx.x;
  // side job: zero the fused attn_merge_o accumulator (runs right before it
  // in stream order; the grid has ~160k threads vs ~10k floats to clear, so
  // the separate fill launch is absorbed for free — same pattern as
  // moe_gemv_h's out_zero)
  if (o_zero != nullptr) {
    const long gid = (((long)blockIdx.z * gridDim.y + blockIdx.y) * gridDim.x
                      + blockIdx.x) * blockDim.x + tid;
    if (gid < zero_n) o_zero[gid] = 0.f;
  }
  const int h = tid >> 5;
  const int sub = tid & 31;
  const int seq = seq_ids[t];
  const int bound = q_pos[t] + 1;
  const int n_qheads = n_kvheads * QH_PER_KV;
  const int hq = hk * QH_PER_KV + h;

  // split range, CH-aligned; split count = launch grid.z (32 short / 64 long
  // contexts), buffer layout stride fixed at NSPLITS_MAX
  const int nsp = gridDim.z;
  const int span = ((bound + nsp - 1) / nsp + CH - 1) & ~(CH - 1);
  const int lo = split * span;
  const int hi = min(bound, lo + span);

  float* ml = part_ml + (((long)t * n_qheads + hq) * NSPLITS_MAX + split) * 2;
  float* acc_out = part + (((long)t * n_qheads + hq) * NSPLITS_MAX + split) * HEAD_DIM;

  if (lo >= hi) {  // empty split still writes a neutral partial
    if (h < QH_PER_KV) {
      if (sub == 0) { ml[0] = -INFINITY; ml[1] = 0.f; }
      *reinterpret_cast<f32x4*>(acc_out + sub * 4) = f32x4{0.f, 0.f, 0.f, 0.f};
    }
    return;
  }

  // K/V chunks are staged through LDS cooperatively (fully-coalesced bulk
  // HBM loads, one read per WG instead of one per q-head); the score and
  // value phases then run out of LDS. Row pad of 8 bf16 keeps the per-lane
  // row reads off a single bank (same layout as flash_prefill).
  #define PA_PAD 8
  constexpr int NBUF = PIPE ? 2 : 1;
  __shared__ float q_s[QH_PER_KV][HEAD_DIM];
  __shared__ float p_s[QH_PER_KV][CH];
  __shared__ short k_s[NBUF][CH][HEAD_DIM + PA_PAD];
  __shared__ short v_s[NBUF][CH][HEAD_DIM + PA_PAD];
  for (int i = tid; i < QH_PER_KV * HEAD_DIM; i += blockDim.x) {
    int hh = i / HEAD_DIM, dd = i % HEAD_DIM;
    q_s[hh][dd] = bf2f(q[(long)t * q_row_stride
                         + (hk * QH_PER_KV + hh) * HEAD_DIM + dd]);
  }

  float acc[4] = {0.f, 0.f, 0.f, 0.f};
  float m_run = -INFINITY, l_run = 0.f;
  const long kv_stride_block = (long)n_kvheads * BLOCK_SIZE * HEAD_DIM;
  const int* btab = block_table + (long)seq * max_blocks;

  // per-thread staging registers (CH/16 vec8 pairs)
  bf16x8 kreg[CH / 16], vreg[CH / 16];
  auto issue_loads = [&](int base) {
    #pragma unroll
    for (int it = 0; it < CH / 16; ++it) {
      const int idx = tid + it * 256;
      const int pos_l = (idx * 8) / HEAD_DIM;
      const int d8 = (idx * 8) % HEAD_DIM;
      const int pos = base + pos_l;
      kreg[it] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      vreg[it] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      if (pos < hi) {
        const int blk = btab[pos / BLOCK_SIZE];
        const long off = (long)blk * kv_stride_block
                         + ((long)hk * BLOCK_SIZE + (pos % BLOCK_SIZE)) * HEAD_DIM + d8;
        kreg[it] = *reinterpret_cast<const bf16x8*>(kcache + off);
        vreg[it] = *reinterpret_cast<const bf16x8*>(vcache + off);
      }
    }
  };
  auto store_regs = [&](int buf) {
    #pragma unroll
    for (int it = 0; it < CH / 16; ++it) {
      const int idx = tid + it * 256;
      const int pos_l = (idx * 8) / HEAD_DIM;
      const int d8 = (idx * 8) % HEAD_DIM;
      *reinterpret_cast<bf16x8*>(&k_s[buf][pos_l][d8]) = kreg[it];
      *reinterpret_cast<bf16x8*>(&v_s[buf][pos_l][d8]) = vreg[it];
    }
  };

  issue_loads(lo);
  int iter = 0;
  for (int base = lo; base < hi; base += CH, ++iter) {
    const int cur = PIPE ? (iter & 1) : 0;
    store_regs(cur);
    __syncthreads();                 // staged chunk visible to all waves
    if (PIPE && base + CH < hi)
      issue_loads(base + CH);        // HBM latency overlaps score+value

    {   // score: thread (h, sub) dots q_s[h] with k_s[cur][sub + r*32]
      #pragma unroll
      for (int r = 0; r < CH / 32; ++r) {
        const int p = sub + r * 32;
        float s = -INFINITY;
        if (base + p < hi) {
          float dot = 0.f;
          #pragma unroll
          for (int v8 = 0; v8 < HEAD_DIM / 8; ++v8) {
            bf16x8 kv = *reinterpret_cast<const bf16x8*>(&k_s[cur][p][v8 * 8]);
            #pragma unroll
            for (int j = 0; j < 8; ++j) dot += q_s[h][v8 * 8 + j] * bf2f(kv[j]);
          }
          s = dot * scale;
        }
        p_s[h][p] = s;
      }
    }
    __syncthreads();
    {
      float chunk_max = -INFINITY;
      #pragma unroll
      for (int j = 0; j < CH; ++j) chunk_max = fmaxf(chunk_max, p_s[h][j]);
      const float m_new = fmaxf(m_run, chunk_max);
      if (m_new != -INFINITY) {
        const float rescale = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
        acc[0] *= rescale; acc[1] *= rescale; acc[2] *= rescale; acc[3] *= rescale;
        l_run *= rescale;
        const int lim = min(CH, hi - base);
        for (int j = 0; j < lim; ++j) {
          const float w = __expf(p_s[h][j] - m_new);
          l_run += w;
          bf16x4 vv = *reinterpret_cast<const bf16x4*>(&v_s[cur][j][sub * 4]);
          acc[0] += w * bf2f(vv[0]);
          acc[1] += w * bf2f(vv[1]);
          acc[2] += w * bf2f(vv[2]);
          acc[3] += w * bf2f(vv[3]);
        }
        m_run = m_new;
      }
    }
    __syncthreads();  // value reads done before the next store_regs
  }

  if (sub == 0) { ml[0] = m_run; ml[1] = l_run; }
  *reinterpret_cast<f32x4*>(acc_out + sub * 4) =
      f32x4{acc[0], acc[1], acc[2], acc[3]};
}

// merge: one wave per (t, hq); lane d-pairs combine the NSPLITS partials
template <int NS>
__global__ __launch_bounds__(128)
void paged_attn_merge_kernel(short* __restrict__ out,       // [T, Hq, D]
                             const float* __restrict__ part,
                             const float* __restrict__ part_ml,
                             int n_qheads) {
  const int t = blockIdx.x;
  const int hq = blockIdx.y;
  const int d = threadIdx.x;  // 128 threads = one dim each
  const float* ml = part_ml + (((long)t * n_qheads + hq) * NSPLITS_MAX) * 2;
  const float* pacc = part + (((long)t * n_qheads + hq) * NSPLITS_MAX) * HEAD_DIM;

  float m_star = -INFINITY;
  #pragma unroll
  for (int s = 0; s < NS; ++s) m_star = fmaxf(m_star, ml[2 * s]);
  float l_tot = 0.f, a = 0.f;
  #pragma unroll
  for (int s = 0; s < NS; ++s) {
    const float ms = ml[2 * s];
    if (ms == -INFINITY) continue;
    const float f = __expf(ms - m_star);
    l_tot += ml[2 * s + 1] * f;
    a += pacc[(long)s * HEAD_DIM + d] * f;
  }
  out[((long)t * n_qheads + hq) * HEAD_DIM + d] =
      f2bf(l_tot > 0.f ? a / l_tot : 0.f);
}

int64_t attn_nsplits() { return NSPLITS_MAX; }

// split count for a given max context: 64 splits pay off past ~8k tokens
// (shorter per-WG serial chunk chains beat the extra merge traffic there —
// round-1 A/B: NSPLITS=64 wins only at L>=16k vs static-32 everywhere; with
// banded decode graphs each band gets its own captured grid)
static int pick_splits(int max_ctx) { return max_ctx >= 8192 ? 64 : NSPLITS; }

void paged_attention_split(torch::Tensor out, torch::Tensor q,
                           torch::Tensor kcache, torch::Tensor vcache,
                           torch::Tensor block_table, torch::Tensor seq_ids,
                           torch::Tensor q_pos, torch::Tensor part,
                           torch::Tensor part_ml, double scale,
                           int64_t splits) {
  TORCH_CHECK(splits == 32 || splits == 64, "splits must be 32 or 64");
  // q may be a strided row-view into the packed qkv buffer
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2));
  const int T = q.size(0);
  const int n_kvheads = kcache.size(1);
  const int n_qheads = n_kvheads * QH_PER_KV;
  const int max_blocks = block_table.size(1);
  TORCH_CHECK(part.size(0) >= T && part.size(2) == NSPLITS_MAX);
  int max_ctx = 0;  // host-known bound: use tensor length heuristic (eager
  // path only; graph path passes splits explicitly via the splitk variant)
  hipStream_t s = c10::hip::getCurrentHIPStream();
  dim3 g1(T, n_kvheads, (unsigned)splits);
  (void)max_ctx;
  #define PA_SPLIT_ARGS part.data_ptr<float>(), part_ml.data_ptr<float>(), \
      (const short*)q.data_ptr(), (const short*)kcache.data_ptr(), \
      (const short*)vcache.data_ptr(), block_table.data_ptr<int>(), \
      seq_ids.data_ptr<int>(), q_pos.data_ptr<int>(), \
      n_kvheads, max_blocks, (int)q.stride(0), (float)scale, \
      (float*)nullptr, 0L
  if (attn_chunk() == 64) {
    if (attn_pipe())
      hipLaunchKernelGGL((paged_attn_split_kernel<64, true>), g1, dim3(256), 0, s, PA_SPLIT_ARGS);
    else
      hipLaunchKernelGGL((paged_attn_split_kernel<64, false>), g1, dim3(256), 0, s, PA_SPLIT_ARGS);
  } else {
    if (attn_pipe())
      hipLaunchKernelGGL((paged_attn_split_kernel<32, true>), g1, dim3(256), 0, s, PA_SPLIT_ARGS);
    else
      hipLaunchKernelGGL((paged_attn_split_kernel<32, false>), g1, dim3(256), 0, s, PA_SPLIT_ARGS);
  }
  #undef PA_SPLIT_ARGS
  HIP_CHECK_KERNEL();
  dim3 g2(T, n_qheads);
  if (splits == 64)
    hipLaunchKernelGGL(paged_attn_merge_kernel<64>, g2, dim3(HEAD_DIM), 0, s,
                       (short*)out.data_ptr(), part.data_ptr<float>(),
                       part_ml.data_ptr<float>(), n_qheads);
  else
    hipLaunchKernelGGL(paged_attn_merge_kernel<32>, g2, dim3(HEAD_DIM), 0, s,
                       (short*)out.data_ptr(), part.data_ptr<float>(),
                       part_ml.data_ptr<float>(), n_qheads);
  HIP_CHECK_KERNEL();
}

// split-only variant for the fused attn_merge_o path: no bf16 merge output;
// zeroes the downstream f32 O accumulator as a side job.
void paged_attention_splitk(torch::Tensor part, torch::Tensor part_ml,
                            torch::Tensor q, torch::Tensor kcache,
                            torch::Tensor vcache, torch::Tensor block_table,
                            torch::Tensor seq_ids, torch::Tensor q_pos,
                            double scale, torch::Tensor o_zero,
                            int64_t splits) {
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2));
  const int T = q.size(0);
  const int n_kvheads = kcache.size(1);
  const int max_blocks = block_table.size(1);
  TORCH_CHECK(part.size(0) >= T && part.size(2) == NSPLITS_MAX);
  TORCH_CHECK(o_zero.dtype() == torch::kFloat32 && o_zero.is_contiguous());
  TORCH_CHECK(splits == 32 || splits == 64, "splits must be 32 or 64");
  hipStream_t s = c10::hip::getCurrentHIPStream();
  dim3 g1(T, n_kvheads, (unsigned)splits);
  #define PA_SPLITK_ARGS part.data_ptr<float>(), part_ml.data_ptr<float>(), \
      (const short*)q.data_ptr(), (const short*)kcache.data_ptr(), \
      (const short*)vcache.data_ptr(), block_table.data_ptr<int>(), \
      seq_ids.data_ptr<int>(), q_pos.data_ptr<int>(), \
      n_kvheads, max_blocks, (int)q.stride(0), (float)scale, \
      o_zero.data_ptr<float>(), (long)o_zero.numel()
  if (attn_chunk() == 64) {
    if (attn_pipe())
      hipLaunchKernelGGL((paged_attn_split_kernel<64, true>), g1, dim3(256), 0, s, PA_SPLITK_ARGS);
    else
      hipLaunchKernelGGL((paged_attn_split_kernel<64, false>), g1, dim3(256), 0, s, PA_SPLITK_ARGS);
  } else {
    if (attn_pipe())
      hipLaunchKernelGGL((paged_attn_split_kernel<32, true>), g1, dim3(256), 0, s, PA_SPLITK_ARGS);
    else
      hipLaunchKernelGGL((paged_attn_split_kernel<32, false>), g1, dim3(256), 0, s, PA_SPLITK_ARGS);
  }
  #undef PA_SPLITK_ARGS
  HIP_CHECK_KERNEL();
}
