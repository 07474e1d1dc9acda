// MFMA flash-attention prefill over the paged KV cache.
//
// Replaces the vector-ALU prefill path (paged_attn_kernel) which profiled at
// 62% of prefill time (3.06 ms/layer @ T=2048 — scripts/gpu_prefill_attrib.py):
// per-thread 128-dim dots re-read K 8× per head group and the value phase
// serialized on L1 latency. This kernel is flash-style: Q-tile × KV-tile with
// online softmax, QK^T and PV on mfma_f32_16x16x32_bf16, K/V staged through
// LDS once per chunk.
//
// Geometry: one workgroup = (q-tile of ≤16 tokens of ONE sequence) × (kv head).
//   512 threads = 8 waves; wave w = q-head w of the GQA group.
//   M = 16 tokens (wave-local), N = 32 kv positions/chunk, D = 128.
//   QK^T: C[16,32] = Q[16,128]·K^T — 2 n-frags × 4 k-steps = 8 MFMA/chunk.
//   PV:   O[16,128] += P[16,32]·V — 8 n-frags × 1 k-step = 8 MFMA/chunk.
// C-fragment mapping (verified): row=(lane>>4)*4+r, col=lane&15.
// LDS rows padded so b128 reads across rows are ≤2-way bank conflicts (free).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

#define FP_BS 16          // paged KV block size
#define FP_QTOK 16        // q tokens per tile
#define FP_CHUNK 32       // kv positions per chunk
#define FP_D 128
#define FP_QH 8           // GQA group
#define FP_PADK 8         // bf16 pad for K/V/P LDS rows

typedef __attribute__((ext_vector_type(4))) float fpfrag_t;

__global__ __launch_bounds__(512, 1)
void flash_prefill_kernel(short* __restrict__ out,         // [T, Hq, D]
                          const short* __restrict__ q,      // [T, Hq, D]
                          const short* __restrict__ kcache, // [NB, Hk, 16, D]
                          const short* __restrict__ vcache,
                          const int* __restrict__ block_table,
                          const int* __restrict__ seq_ids,
                          const int* __restrict__ q_pos,
                          const int* __restrict__ tile_desc,  // [G, 2]: row0, n
                          int n_kvheads, int max_blocks, int q_row_stride,
                          float scale) {
  const int g = blockIdx.x;
  const int hk = blockIdx.y;
  const int row0 = tile_desc[g * 2 + 0];
  const int ntok = tile_desc[g * 2 + 1];
  const int tid = threadIdx.x;
  const int wid = tid >> 6;       // q head within group
  const int lane = tid & 63;
  const int n_qheads = n_kvheads * FP_QH;
  const int hq = hk * FP_QH + wid;

  __shared__ short q_s[FP_QH][FP_QTOK][FP_D + FP_PADK];
  __shared__ short k_s[FP_CHUNK][FP_D + FP_PADK];
  __shared__ short vt_s[FP_D][FP_CHUNK + FP_PADK];
  __shared__ short p_s[FP_QH][FP_QTOK][FP_CHUNK + FP_PADK];
  __shared__ int qp_s[FP_QTOK];
  __shared__ int seq_s;

  // ---- load Q tile: 8h × 16tok × 128d = 2048 vec8 → 4 per thread
  #pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int idx = tid + it * 512;           // vec8 index
    const int h = idx / (FP_QTOK * FP_D / 8);
    const int rem = idx % (FP_QTOK * FP_D / 8);
    const int tk = rem / (FP_D / 8);
    const int d8 = (rem % (FP_D / 8)) * 8;
    bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (tk < ntok)
      v = *reinterpret_cast<const bf16x8*>(
          q + (long)(row0 + tk) * q_row_stride + (hk * FP_QH + h) * FP_D + d8);
    *reinterpret_cast<bf16x8*>(&q_s[h][tk][d8]) = v;
  }
  if (tid < FP_QTOK)
    qp_s[tid] = (tid < ntok) ? q_pos[row0 + tid] : -1;
  if (tid == 0) seq_s = seq_ids[row0];
  __syncthreads();

  const int seq = seq_s;
  int bound_max = 0;
  for (int i = 0; i < ntok; ++i) bound_max = max(bound_max, qp_s[i] + 1);
  const int* btab = block_table + (long)seq * max_blocks;
  const long kv_stride_block = (long)n_kvheads * FP_BS * FP_D;

  // per-lane row stats: rows (lane>>4)*4 + r
  float m_run[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
  float l_run[4] = {0.f, 0.f, 0.f, 0.f};
  fpfrag_t oacc[8];
  #pragma unroll
  for (int nf = 0; nf < 8; ++nf) oacc[nf] = fpfrag_t{0.f, 0.f, 0.f, 0.f};

  for (int base = 0; base < bound_max; base += FP_CHUNK) {
    // ---- stage K chunk [32][128] and V^T [128][32] cooperatively
    // 32×128/8 = 512 vec8: one per thread
    {
      const int pos_l = (tid * 8) / FP_D;      // 0..31
      const int d8 = (tid * 8) % FP_D;
      const int pos = base + pos_l;
      bf16x8 kv = {0, 0, 0, 0, 0, 0, 0, 0};
      bf16x8 vv = {0, 0, 0, 0, 0, 0, 0, 0};
      if (pos < bound_max) {
        const int blk = btab[pos / FP_BS];
        const long off = (long)blk * kv_stride_block
                         + ((long)hk * FP_BS + (pos % FP_BS)) * FP_D + d8;
        kv = *reinterpret_cast<const bf16x8*>(kcache + off);
        vv = *reinterpret_cast<const bf16x8*>(vcache + off);
      }
      *reinterpret_cast<bf16x8*>(&k_s[pos_l][d8]) = kv;
      #pragma unroll
      for (int j = 0; j < 8; ++j) vt_s[d8 + j][pos_l] = vv[j];
    }
    __syncthreads();

    // ---- QK^T: C[16,32] per wave
    fpfrag_t sfrag[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
    #pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const int akoff = kk * 32 + (lane >> 4) * 8;
      bf16x8 a = *reinterpret_cast<const bf16x8*>(&q_s[wid][lane & 15][akoff]);
      #pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &k_s[nf * 16 + (lane & 15)][akoff]);
        sfrag[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, sfrag[nf], 0, 0, 0);
      }
    }

    // ---- causal mask + online softmax (per row r: token (lane>>4)*4+r)
    float p_vals[2][4];
    float rescale[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int tk = (lane >> 4) * 4 + r;
      const int bound = qp_s[tk] + 1;  // -1+1=0 for pad rows → all masked
      float s0 = sfrag[0][r] * scale;
      float s1 = sfrag[1][r] * scale;
      const int pos0 = base + (lane & 15);
      const int pos1 = pos0 + 16;
      if (pos0 >= bound) s0 = -INFINITY;
      if (pos1 >= bound) s1 = -INFINITY;
      // row max across the 16-lane group (cols)
      float mx = fmaxf(s0, s1);
      #pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
      const float m_new = fmaxf(m_run[r], mx);
      float p0 = 0.f, p1 = 0.f, rs = 1.f;
      if (m_new != -INFINITY) {
        rs = (m_run[r] == -INFINITY) ? 1.f : __expf(m_run[r] - m_new);
        p0 = (s0 == -INFINITY) ? 0.f : __expf(s0 - m_new);
        p1 = (s1 == -INFINITY) ? 0.f : __expf(s1 - m_new);
        float lsum = p0 + p1;
        #pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          lsum += __shfl_xor(lsum, off, WAVE);
        l_run[r] = l_run[r] * rs + lsum;
        m_run[r] = m_new;
      }
      rescale[r] = rs;
      p_vals[0][r] = p0;
      p_vals[1][r] = p1;
    }

    // ---- write P to LDS (transpose C-layout → A-layout)
    #pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int tk = (lane >> 4) * 4 + r;
        p_s[wid][tk][nf * 16 + (lane & 15)] = f2bf(p_vals[nf][r]);
      }
    }
    // rescale O accumulators (row r factor applies to oacc[*][r])
    #pragma unroll
    for (int nf = 0; nf < 8; ++nf) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) oacc[nf][r] *= rescale[r];
    }
    __syncthreads();  // P + Vt visible

    // ---- PV: O[16,128] += P[16,32] · V[32,128]
    {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &p_s[wid][lane & 15][(lane >> 4) * 8]);
      #pragma unroll
      for (int nf = 0; nf < 8; ++nf) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &vt_s[nf * 16 + (lane & 15)][(lane >> 4) * 8]);
        oacc[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, oacc[nf], 0, 0, 0);
      }
    }
    __syncthreads();  // before restaging K/V
  }

  // ---- epilogue: normalize and store
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int tk = (lane >> 4) * 4 + r;
    if (tk >= ntok) continue;
    const float inv_l = (l_run[r] > 0.f) ? 1.0f / l_run[r] : 0.f;
    #pragma unroll
    for (int nf = 0; nf < 8; ++nf) {
      out[((long)(row0 + tk) * n_qheads + hq) * FP_D + nf * 16 + (lane & 15)] =
          f2bf(oacc[nf][r] * inv_l);
    }
  }
}

void flash_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor kcache,
                   torch::Tensor vcache, torch::Tensor block_table,
                   torch::Tensor seq_ids, torch::Tensor q_pos,
                   torch::Tensor tile_desc, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(q.size(-1) == FP_D);
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == FP_D);  // strided row-views OK
  const int n_kvheads = kcache.size(1);
  TORCH_CHECK(q.size(1) == n_kvheads * FP_QH);
  const int G = tile_desc.size(0);
  const int max_blocks = block_table.size(1);
  dim3 grid(G, n_kvheads), block(512);
  hipStream_t s = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(flash_prefill_kernel, grid, block, 0, s,
                     (short*)out.data_ptr(), (const short*)q.data_ptr(),
                     (const short*)kcache.data_ptr(),
                     (const short*)vcache.data_ptr(),
                     block_table.data_ptr<int>(), seq_ids.data_ptr<int>(),
                     q_pos.data_ptr<int>(), tile_desc.data_ptr<int>(),
                     n_kvheads, max_blocks, (int)q.stride(0), (float)scale);
  HIP_CHECK_KERNEL();
}
