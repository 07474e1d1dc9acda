// MFMA flash-attention prefill over the paged KV cache.
//
// Replaces the vector-ALU prefill path (paged_attn_kernel) which profiled at
// 62% of prefill time (3.06 ms/layer @ T=2048 — scripts/gpu_prefill_attrib.py).
// Flash-style: Q-tile × KV-chunk with online softmax, QK^T and PV on
// mfma_f32_16x16x32_bf16, K/V staged through LDS once per chunk.
//
// Round-2 geometry (v1 was M=16/N=32 and measured MFMA:VALU 1:16 — the
// softmax bookkeeping drowned the matrix pipe): one workgroup =
// (q-tile of ≤32 tokens of ONE sequence) × (kv head); 512 threads = 8 waves;
// wave w = q-head w of the GQA group.
//   M = 32 tokens (2 row-fragments), N = 64 kv positions/chunk (4 col-frags),
//   D = 128. Per chunk per wave: QK^T 2m×4n×4k = 32 MFMA, PV 2m×8n×2k = 32.
//   4× the MFMA work per staged chunk of v1, while the cross-lane max/sum
//   reductions stay one pair per row per chunk (amortized over 4 col-frags)
//   and the 3 block barriers cover 2048 scores instead of 512.
// C-fragment mapping (verified): row=(lane>>4)*4+r, col=lane&15.
// LDS rows padded so b128 reads across rows are ≤2-way bank conflicts (free).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

#define FP_BS 16          // paged KV block size
#define FP_QTOK 32        // q tokens per tile (2 MFMA row-blocks)
#define FP_CHUNK 64       // kv positions per chunk (4 MFMA col-blocks)
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

  __shared__ short q_s[FP_QH][FP_QTOK][FP_D + FP_PADK];       // 69.6 KB
  __shared__ short k_s[FP_CHUNK][FP_D + FP_PADK];             // 17.4 KB
  __shared__ short vt_s[FP_D][FP_CHUNK + FP_PADK];            // 18.4 KB
  __shared__ short p_s[FP_QH][FP_QTOK][FP_CHUNK + FP_PADK];   // 36.9 KB
  __shared__ int qp_s[FP_QTOK];
  __shared__ int seq_s;

  // ---- load Q tile: 8h × 32tok × 128d = 4096 vec8 → 8 per thread
  #pragma unroll
  for (int it = 0; it < 8; ++it) {
    const int idx = tid + it * 512;           // vec8 index
    const int h = idx / (FP_QTOK * FP_D / 8);
    const int rem = idx % (FP_QTOK * FP_D / 8);
    const int tk = rem / (FP_D / 8);
    const int d8 = (rem % (FP_D / 8)) * 8;
    bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (tk < ntok) {
      v = *reinterpret_cast<const bf16x8*>(
          q + (long)(row0 + tk) * q_row_stride + (hk * FP_QH + h) * FP_D + d8);
      // fold the softmax scale into Q here: saves one VALU mul per score
      // in the hot per-chunk loop (8 per lane per chunk)
      #pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = f2bf(bf2f(v[j]) * scale);
    }
    *reinterpret_cast<bf16x8*>(&q_s[h][tk][d8]) = v;
  }
  if (tid < FP_QTOK)
    qp_s[tid] = (tid < ntok) ? q_pos[row0 + tid] : -1;
  if (tid == 0) seq_s = seq_ids[row0];
  __syncthreads();

  const int seq = seq_s;
  int bound_max = 0, bound_min = INT_MAX;
  for (int i = 0; i < ntok; ++i) {
    bound_max = max(bound_max, qp_s[i] + 1);
    bound_min = min(bound_min, qp_s[i] + 1);
  }
  const int* btab = block_table + (long)seq * max_blocks;
  const long kv_stride_block = (long)n_kvheads * FP_BS * FP_D;

  // per-lane row stats: m-tile mt rows mt*16 + (lane>>4)*4 + r
  float m_run[2][4], l_run[2][4];
  fpfrag_t oacc[2][8];
  #pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) { m_run[mt][r] = -INFINITY; l_run[mt][r] = 0.f; }
    #pragma unroll
    for (int nf = 0; nf < 8; ++nf) oacc[mt][nf] = fpfrag_t{0.f, 0.f, 0.f, 0.f};
  }

  for (int base = 0; base < bound_max; base += FP_CHUNK) {
    // ---- stage K chunk [64][128] and V^T [128][64]: 1024 vec8 → 2/thread
    #pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int idx = tid + it * 512;
      const int pos_l = (idx * 8) / FP_D;      // 0..63
      const int d8 = (idx * 8) % FP_D;
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

    // ---- QK^T: C[32,64] per wave = 2 m-tiles × 4 n-tiles × 4 k-steps
    fpfrag_t sfrag[2][4];
    #pragma unroll
    for (int mt = 0; mt < 2; ++mt)
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt)
        sfrag[mt][nt] = fpfrag_t{0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const int akoff = kk * 32 + (lane >> 4) * 8;
      bf16x8 a0 = *reinterpret_cast<const bf16x8*>(&q_s[wid][lane & 15][akoff]);
      bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
          &q_s[wid][16 + (lane & 15)][akoff]);
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &k_s[nt * 16 + (lane & 15)][akoff]);
        sfrag[0][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b, sfrag[0][nt], 0, 0, 0);
        sfrag[1][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b, sfrag[1][nt], 0, 0, 0);
      }
    }

    // ---- causal mask + online softmax; one max/sum lane-reduce pair per
    // row per chunk, amortized over the 4 col-frags. Chunks entirely below
    // every row's causal bound (the common case for full tiles deep in the
    // context) skip the per-score mask compares.
    const int colL = lane & 15;
    const bool interior = (ntok == FP_QTOK) && (base + FP_CHUNK <= bound_min);
    #pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int tk = mt * 16 + (lane >> 4) * 4 + r;
        const int bound = qp_s[tk] + 1;  // -1+1=0 for pad rows → all masked
        float sv[4];
        float mx = -INFINITY;
        if (interior) {
          #pragma unroll
          for (int nt = 0; nt < 4; ++nt) {
            sv[nt] = sfrag[mt][nt][r];   // scale folded into Q at staging
            mx = fmaxf(mx, sv[nt]);
          }
        } else {
          #pragma unroll
          for (int nt = 0; nt < 4; ++nt) {
            float s = sfrag[mt][nt][r];
            if (base + nt * 16 + colL >= bound) s = -INFINITY;
            sv[nt] = s;
            mx = fmaxf(mx, s);
          }
        }
        #pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
        const float m_new = fmaxf(m_run[mt][r], mx);
        float rs = 1.f, lsum = 0.f;
        float pv[4] = {0.f, 0.f, 0.f, 0.f};
        if (m_new != -INFINITY) {
          rs = (m_run[mt][r] == -INFINITY) ? 1.f : __expf(m_run[mt][r] - m_new);
          #pragma unroll
          for (int nt = 0; nt < 4; ++nt) {
            pv[nt] = (sv[nt] == -INFINITY) ? 0.f : __expf(sv[nt] - m_new);
            lsum += pv[nt];
          }
          #pragma unroll
          for (int off = 8; off > 0; off >>= 1)
            lsum += __shfl_xor(lsum, off, WAVE);
          l_run[mt][r] = l_run[mt][r] * rs + lsum;
          m_run[mt][r] = m_new;
        }
        // write P row to LDS (C-layout → A-layout) + rescale O row.
        // rs==1 is the common case once the running max stabilizes; the rs
        // check is row-uniform across the 16 col lanes, so divergence cost
        // stays within the wave's row groups.
        #pragma unroll
        for (int nt = 0; nt < 4; ++nt)
          p_s[wid][tk][nt * 16 + colL] = f2bf(pv[nt]);
        if (rs != 1.f) {
          #pragma unroll
          for (int nf = 0; nf < 8; ++nf) oacc[mt][nf][r] *= rs;
        }
      }
    }
    __syncthreads();  // P + Vt visible

    // ---- PV: O[32,128] += P[32,64] · V[64,128]
    #pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int akoff = kk * 32 + (lane >> 4) * 8;
      bf16x8 a0 = *reinterpret_cast<const bf16x8*>(&p_s[wid][lane & 15][akoff]);
      bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
          &p_s[wid][16 + (lane & 15)][akoff]);
      #pragma unroll
      for (int nf = 0; nf < 8; ++nf) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &vt_s[nf * 16 + (lane & 15)][akoff]);
        oacc[0][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b, oacc[0][nf], 0, 0, 0);
        oacc[1][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b, oacc[1][nf], 0, 0, 0);
      }
    }
    __syncthreads();  // before restaging K/V
  }

  // ---- epilogue: normalize and store
  #pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int tk = mt * 16 + (lane >> 4) * 4 + r;
      if (tk >= ntok) continue;
      const float inv_l = (l_run[mt][r] > 0.f) ? 1.0f / l_run[mt][r] : 0.f;
      #pragma unroll
      for (int nf = 0; nf < 8; ++nf) {
        out[((long)(row0 + tk) * n_qheads + hq) * FP_D + nf * 16 + (lane & 15)] =
            f2bf(oacc[mt][nf][r] * inv_l);
      }
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

int64_t flash_prefill_qtile() { return FP_QTOK; }
