// Paged-KV attention kernels (decode + chunked prefill) for MI355X.
//
// v1 structure: correctness-first VALU kernels with online softmax, paged KV
// gather, GQA sharing (decode reads each K/V page once per KV head and serves
// all its query heads), fp32 accumulation, bf16 storage.  The MFMA-tiled
// prefill (guide §5.5 T10/T14/T16 structure) replaces phase A when profiling
// shows prefill attention on the critical path; decode is HBM-bound on KV
// reads, which this layout already streams.
//
// KV cache layout: [num_blocks, Hkv, BLOCK_SIZE, D] bf16 — a (block, head)
// panel is BLOCK_SIZE*D*2 bytes contiguous, so both phases read contiguous
// rows.

#include "common.h"

#define MAX_GQ 8      // query heads per KV head (llama3-8b: 4)
#define CHUNK 128     // KV tokens processed per software chunk
#define DECODE_BLOCK 128

// ---------------------------------------------------------------------------
// paged_attn_decode: one query token per sequence.
//   q:    [B, Hq, D] bf16      out: [B, Hq, D] bf16
//   kc/vc:[NB, Hkv, BS, D] bf16
//   bt:   [B, MAXB] int32 block tables;  ctx: [B] int32 context lengths
// Grid: (B, Hkv); block: DECODE_BLOCK threads (D <= DECODE_BLOCK).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(DECODE_BLOCK)
paged_attn_decode_kernel(bf16 *__restrict__ out, const bf16 *__restrict__ q,
                         const bf16 *__restrict__ kc,
                         const bf16 *__restrict__ vc,
                         const int *__restrict__ bt,
                         const int *__restrict__ ctx,
                         float scale, int Hq, int Hkv, int D, int BS,
                         int MAXB, int GQ) {
  const int b = blockIdx.x;
  const int hk = blockIdx.y;
  const int tid = threadIdx.x;
  const int len = ctx[b];
  if (len <= 0) return;

  __shared__ float q_s[MAX_GQ * 128];
  __shared__ float p_s[MAX_GQ][CHUNK];
  __shared__ float scratch[8];

  // Load the GQ query heads that share this KV head into LDS (fp32).
  for (int i = tid; i < GQ * D; i += blockDim.x) {
    int g = i / D, d = i % D;
    q_s[g * D + d] = bf2f(q[((long)b * Hq + hk * GQ + g) * D + d]) * scale;
  }
  __syncthreads();

  float m[MAX_GQ], l[MAX_GQ], acc[MAX_GQ];
#pragma unroll
  for (int g = 0; g < MAX_GQ; ++g) {
    m[g] = -INFINITY;
    l[g] = 0.f;
    acc[g] = 0.f;
  }

  const long panel_stride = (long)Hkv * BS * D;  // one cache block
  for (int start = 0; start < len; start += CHUNK) {
    const int clen = min(CHUNK, len - start);

    // Phase A: thread -> KV token; dot against all GQ query heads.
    if (tid < clen) {
      const int token = start + tid;
      const long blk = bt[(long)b * MAXB + token / BS];
      const bf16 *krow =
          kc + blk * panel_stride + ((long)hk * BS + token % BS) * D;
      float dots[MAX_GQ];
#pragma unroll
      for (int g = 0; g < MAX_GQ; ++g) dots[g] = 0.f;
      for (int d8 = 0; d8 < D / 8; ++d8) {
        uint4 kv = reinterpret_cast<const uint4 *>(krow)[d8];
        float kf[8];
        unpack_bf16x2(kv.x, kf[0], kf[1]);
        unpack_bf16x2(kv.y, kf[2], kf[3]);
        unpack_bf16x2(kv.z, kf[4], kf[5]);
        unpack_bf16x2(kv.w, kf[6], kf[7]);
        // static bounds + guard keep dots[] in registers (guide §5.4 rule 20)
#pragma unroll
        for (int g = 0; g < MAX_GQ; ++g) {
          if (g >= GQ) break;
          const float *qg = q_s + g * D + d8 * 8;
#pragma unroll
          for (int k = 0; k < 8; ++k) dots[g] = fmaf(kf[k], qg[k], dots[g]);
        }
      }
#pragma unroll
      for (int g = 0; g < MAX_GQ; ++g) {
        if (g >= GQ) break;
        p_s[g][tid] = dots[g];
      }
    }
    __syncthreads();

    // Online-softmax update per query head (all threads compute identical
    // reductions; block_sum/max return the value to every thread).
    for (int g = 0; g < GQ; ++g) {
      float mine = (tid < clen) ? p_s[g][tid] : -INFINITY;
      float cmax = block_max(mine, scratch);
      float mn = fmaxf(m[g], cmax);
      float alpha = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - mn);
      float p = (tid < clen) ? __expf(p_s[g][tid] - mn) : 0.f;
      if (tid < clen) p_s[g][tid] = p;
      float psum = block_sum(p, scratch);
      l[g] = l[g] * alpha + psum;
      acc[g] *= alpha;
      m[g] = mn;
    }
    __syncthreads();

    // Phase B: thread -> output dim; accumulate P·V over the chunk.
    if (tid < D) {
      for (int i = 0; i < clen; ++i) {
        const int token = start + i;
        const long blk = bt[(long)b * MAXB + token / BS];
        const bf16 *vrow =
            vc + blk * panel_stride + ((long)hk * BS + token % BS) * D;
        const float v = bf2f(vrow[tid]);
#pragma unroll
        for (int g = 0; g < MAX_GQ; ++g) {
          if (g >= GQ) break;
          acc[g] = fmaf(p_s[g][i], v, acc[g]);
        }
      }
    }
    __syncthreads();
  }

  if (tid < D) {
#pragma unroll
    for (int g = 0; g < MAX_GQ; ++g) {
      if (g >= GQ) break;
      const float o = (l[g] > 0.f) ? acc[g] / l[g] : 0.f;
      out[((long)b * Hq + hk * GQ + g) * D + tid] = f2bf(o);
    }
  }
}

// ---------------------------------------------------------------------------
// paged_attn_prefill: chunked causal attention for new (suffix) tokens over
// the full paged context (cached prefix + the suffix itself, already
// appended to the cache by kv_append).
//
// Work decomposition: one block per (tile, qhead); a tile is QT=16
// consecutive query tokens of one sequence.
//   q:        [Tq, Hq, D] bf16 (all new tokens of the batch, seq-major)
//   out:      [Tq, Hq, D] bf16
//   tile_q0:  [ntiles] int32 — row index of the tile's first token in q
//   tile_qn:  [ntiles] int32 — number of q tokens in this tile (<= 16)
//   tile_seq: [ntiles] int32 — sequence index (selects block table)
//   tile_pos0:[ntiles] int32 — absolute position of the tile's first token
// Grid: (ntiles, Hq); block: 128.
// ---------------------------------------------------------------------------
#define QT 16

extern "C" __global__ void __launch_bounds__(DECODE_BLOCK)
paged_attn_prefill_kernel(bf16 *__restrict__ out, const bf16 *__restrict__ q,
                          const bf16 *__restrict__ kc,
                          const bf16 *__restrict__ vc,
                          const int *__restrict__ bt,
                          const int *__restrict__ tile_q0,
                          const int *__restrict__ tile_qn,
                          const int *__restrict__ tile_seq,
                          const int *__restrict__ tile_pos0,
                          float scale, int Hq, int Hkv, int D, int BS,
                          int MAXB, int GQ) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int hk = h / GQ;
  const int tid = threadIdx.x;
  const int q0 = tile_q0[tile];
  const int qn = tile_qn[tile];
  const int seq = tile_seq[tile];
  const int pos0 = tile_pos0[tile];

  __shared__ float q_s[QT * 128];
  __shared__ float p_s[QT][CHUNK];
  __shared__ float scratch[8];

  for (int i = tid; i < qn * D; i += blockDim.x) {
    int qi = i / D, d = i % D;
    q_s[qi * D + d] = bf2f(q[((long)(q0 + qi) * Hq + h) * D + d]) * scale;
  }
  __syncthreads();

  float m[QT], l[QT], acc[QT];
#pragma unroll
  for (int qi = 0; qi < QT; ++qi) {
    m[qi] = -INFINITY;
    l[qi] = 0.f;
    acc[qi] = 0.f;
  }

  const int kv_limit = pos0 + qn;  // last q row attends to positions < pos0+qn
  const long panel_stride = (long)Hkv * BS * D;

  for (int start = 0; start < kv_limit; start += CHUNK) {
    const int clen = min(CHUNK, kv_limit - start);

    if (tid < clen) {
      const int token = start + tid;
      const long blk = bt[(long)seq * MAXB + token / BS];
      const bf16 *krow =
          kc + blk * panel_stride + ((long)hk * BS + token % BS) * D;
      float dots[QT];
#pragma unroll
      for (int qi = 0; qi < QT; ++qi) dots[qi] = 0.f;
      for (int d8 = 0; d8 < D / 8; ++d8) {
        uint4 kv = reinterpret_cast<const uint4 *>(krow)[d8];
        float kf[8];
        unpack_bf16x2(kv.x, kf[0], kf[1]);
        unpack_bf16x2(kv.y, kf[2], kf[3]);
        unpack_bf16x2(kv.z, kf[4], kf[5]);
        unpack_bf16x2(kv.w, kf[6], kf[7]);
#pragma unroll
        for (int qi = 0; qi < QT; ++qi) {
          if (qi >= qn) break;
          const float *qg = q_s + qi * D + d8 * 8;
#pragma unroll
          for (int k = 0; k < 8; ++k)
            dots[qi] = fmaf(kf[k], qg[k], dots[qi]);
        }
      }
#pragma unroll
      for (int qi = 0; qi < QT; ++qi) {
        if (qi >= qn) break;
        // causal mask: q row qi has absolute position pos0 + qi
        p_s[qi][tid] = (token <= pos0 + qi) ? dots[qi] : -INFINITY;
      }
    }
    __syncthreads();

#pragma unroll
    for (int qi = 0; qi < QT; ++qi) {
      if (qi >= qn) break;
      float mine = (tid < clen) ? p_s[qi][tid] : -INFINITY;
      float cmax = block_max(mine, scratch);
      if (cmax == -INFINITY) {
        // fully masked chunk for this row: zero P so phase B adds nothing
        if (tid < clen) p_s[qi][tid] = 0.f;
        __syncthreads();
        continue;
      }
      float mn = fmaxf(m[qi], cmax);
      float alpha = (m[qi] == -INFINITY) ? 0.f : __expf(m[qi] - mn);
      float p = (tid < clen && p_s[qi][tid] != -INFINITY)
                    ? __expf(p_s[qi][tid] - mn) : 0.f;
      if (tid < clen) p_s[qi][tid] = p;
      float psum = block_sum(p, scratch);
      l[qi] = l[qi] * alpha + psum;
      acc[qi] *= alpha;
      m[qi] = mn;
    }
    __syncthreads();

    if (tid < D) {
      for (int i = 0; i < clen; ++i) {
        const int token = start + i;
        const long blk = bt[(long)seq * MAXB + token / BS];
        const bf16 *vrow =
            vc + blk * panel_stride + ((long)hk * BS + token % BS) * D;
        const float v = bf2f(vrow[tid]);
#pragma unroll
        for (int qi = 0; qi < QT; ++qi) {
          if (qi >= qn) break;
          acc[qi] = fmaf(p_s[qi][i], v, acc[qi]);
        }
      }
    }
    __syncthreads();
  }

  if (tid < D) {
#pragma unroll
    for (int qi = 0; qi < QT; ++qi) {
      if (qi >= qn) break;
      const float o = (l[qi] > 0.f) ? acc[qi] / l[qi] : 0.f;
      out[((long)(q0 + qi) * Hq + h) * D + tid] = f2bf(o);
    }
  }
}

// ---------------------------------------------------------------------------
// Flash-decoding split kernels: the single-pass decode kernel launches only
// B*Hkv workgroups — at agent-pool batch sizes (B<=16) that is <6% of the
// 256 CUs and the context walk serializes.  Split the context across NSPLIT
// workgroups per (seq, kv-head), each producing partial online-softmax state
// (m, l, acc[D]) to a workspace, then combine per query head.
//
//   part_m/part_l: [B, Hq, NS] f32
//   part_acc:      [B, Hq, NS, D] f32
// Grid: (B, Hkv, NS); block: DECODE_BLOCK.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(DECODE_BLOCK)
paged_attn_decode_split_kernel(
    float *__restrict__ part_m, float *__restrict__ part_l,
    float *__restrict__ part_acc, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ ctx, float scale,
    int Hq, int Hkv, int D, int BS, int MAXB, int GQ, int NS) {
  const int b = blockIdx.x;
  const int hk = blockIdx.y;
  const int split = blockIdx.z;
  const int tid = threadIdx.x;
  const int len = ctx[b];
  // contiguous token range for this split (CHUNK-aligned)
  const int chunks = (len + CHUNK - 1) / CHUNK;
  const int per = (chunks + NS - 1) / NS;
  const int t0 = split * per * CHUNK;
  const int t1 = min(len, (split + 1) * per * CHUNK);
  const bool dead = (t0 >= len);

  // K staged through LDS: the direct per-thread row walk gathers 64 lanes
  // from 64 different 256B rows (uncoalesced — profiled at 13% of HBM
  // roofline); the staging loop below streams the same bytes coalesced.
  // Row stride 136 keeps 16B alignment and spreads banks.
#define KSTRIDE 136
  __shared__ float q_s[MAX_GQ * 128];
  __shared__ float p_s[MAX_GQ][CHUNK];
  __shared__ bf16 k_s[CHUNK * KSTRIDE];
  __shared__ float scratch[8];

  for (int i = tid; i < GQ * D; i += blockDim.x) {
    int g = i / D, d = i % D;
    q_s[g * D + d] = bf2f(q[((long)b * Hq + hk * GQ + g) * D + d]) * scale;
  }
  __syncthreads();

  float m[MAX_GQ], l[MAX_GQ], acc[MAX_GQ];
#pragma unroll
  for (int g = 0; g < MAX_GQ; ++g) {
    m[g] = -INFINITY;
    l[g] = 0.f;
    acc[g] = 0.f;
  }

  const long panel_stride = (long)Hkv * BS * D;
  const int dvecs = D / 8;
  for (int start = t0; start < t1; start += CHUNK) {
    const int clen = min(CHUNK, t1 - start);
    for (int i = tid; i < clen * dvecs; i += blockDim.x) {
      const int key = i / dvecs, d8 = i % dvecs;
      const int token = start + key;
      const long blk = bt[(long)b * MAXB + token / BS];
      const uint4 kv = reinterpret_cast<const uint4 *>(
          kc + blk * panel_stride + ((long)hk * BS + token % BS) * D)[d8];
      reinterpret_cast<uint4 *>(k_s + key * KSTRIDE + d8 * 8)[0] = kv;
    }
    __syncthreads();

    if (tid < clen) {
      float dots[MAX_GQ];
#pragma unroll
      for (int g = 0; g < MAX_GQ; ++g) dots[g] = 0.f;
      const bf16 *krow = k_s + tid * KSTRIDE;
      for (int d8 = 0; d8 < dvecs; ++d8) {
        uint4 kv = reinterpret_cast<const uint4 *>(krow)[d8];
        float kf[8];
        unpack_bf16x2(kv.x, kf[0], kf[1]);
        unpack_bf16x2(kv.y, kf[2], kf[3]);
        unpack_bf16x2(kv.z, kf[4], kf[5]);
        unpack_bf16x2(kv.w, kf[6], kf[7]);
#pragma unroll
        for (int g = 0; g < MAX_GQ; ++g) {
          if (g >= GQ) break;
          const float *qg = q_s + g * D + d8 * 8;
#pragma unroll
          for (int k = 0; k < 8; ++k) dots[g] = fmaf(kf[k], qg[k], dots[g]);
        }
      }
#pragma unroll
      for (int g = 0; g < MAX_GQ; ++g) {
        if (g >= GQ) break;
        p_s[g][tid] = dots[g];
      }
    }
    __syncthreads();

    for (int g = 0; g < GQ; ++g) {
      float mine = (tid < clen) ? p_s[g][tid] : -INFINITY;
      float cmax = block_max(mine, scratch);
      float mn = fmaxf(m[g], cmax);
      float alpha = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - mn);
      float p = (tid < clen) ? __expf(p_s[g][tid] - mn) : 0.f;
      if (tid < clen) p_s[g][tid] = p;
      float psum = block_sum(p, scratch);
      l[g] = l[g] * alpha + psum;
      acc[g] *= alpha;
      m[g] = mn;
    }
    __syncthreads();

    // Phase B reads V from LDS, time-sharing k_s (K is fully consumed in
    // phase A): one coalesced bulk load replaces 128 scattered row reads
    // per chunk that measured 6% of HBM peak.
    for (int i = tid; i < clen * dvecs; i += blockDim.x) {
      const int key = i / dvecs, d8 = i % dvecs;
      const int token = start + key;
      const long blk = bt[(long)b * MAXB + token / BS];
      const uint4 vv = reinterpret_cast<const uint4 *>(
          vc + blk * panel_stride + ((long)hk * BS + token % BS) * D)[d8];
      reinterpret_cast<uint4 *>(k_s + key * KSTRIDE + d8 * 8)[0] = vv;
    }
    __syncthreads();

    if (tid < D) {
      for (int i = 0; i < clen; ++i) {
        const float v = bf2f(k_s[i * KSTRIDE + tid]);
#pragma unroll
        for (int g = 0; g < MAX_GQ; ++g) {
          if (g >= GQ) break;
          acc[g] = fmaf(p_s[g][i], v, acc[g]);
        }
      }
    }
    __syncthreads();
  }

  if (tid < D) {
#pragma unroll
    for (int g = 0; g < MAX_GQ; ++g) {
      if (g >= GQ) break;
      const int h = hk * GQ + g;
      const long base = ((long)b * Hq + h) * NS + split;
      if (tid == 0) {
        part_m[base] = dead ? -INFINITY : m[g];
        part_l[base] = dead ? 0.f : l[g];
      }
      part_acc[base * D + tid] = dead ? 0.f : acc[g];
    }
  }
}

// Combine the split partials: out[b,h,:] = sum_i acc_i*exp(m_i-M) / L.
// Grid: (B, Hq); block: DECODE_BLOCK (>= D).
extern "C" __global__ void __launch_bounds__(DECODE_BLOCK)
paged_attn_decode_reduce_kernel(bf16 *__restrict__ out,
                                const float *__restrict__ part_m,
                                const float *__restrict__ part_l,
                                const float *__restrict__ part_acc,
                                int Hq, int D, int NS) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int tid = threadIdx.x;
  const long base = ((long)b * Hq + h) * NS;

  float M = -INFINITY;
  for (int i = tid; i < NS; i += blockDim.x)
    M = fmaxf(M, part_m[base + i]);
  __shared__ float scratch[8];
  M = block_max(M, scratch);

  float L = 0.f;
  for (int i = tid; i < NS; i += blockDim.x) {
    float mi = part_m[base + i];
    L += (mi == -INFINITY) ? 0.f : part_l[base + i] * __expf(mi - M);
  }
  L = block_sum(L, scratch);

  if (tid < D) {
    float o = 0.f;
    for (int i = 0; i < NS; ++i) {
      float mi = part_m[base + i];
      if (mi == -INFINITY) continue;
      o = fmaf(part_acc[(base + i) * D + tid], __expf(mi - M), o);
    }
    out[((long)b * Hq + h) * D + tid] = f2bf(L > 0.f ? o / L : 0.f);
  }
}

// ---------------------------------------------------------------------------
// Split prefill: small late-conversation chunks (grammar forced-runs, few
// tiles) leave the (ntiles, Hq) grid nearly empty while every tile walks the
// WHOLE cached context serially.  Partition the KV walk across NS workgroups
// per (tile, head) exactly like the decode split path.
//   part_m/part_l: [ntiles, Hq, NS, QT] f32
//   part_acc:      [ntiles, Hq, NS, QT, D] f32
// Grid: (ntiles, Hq, NS); block: DECODE_BLOCK.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(DECODE_BLOCK)
paged_attn_prefill_split_kernel(
    float *__restrict__ part_m, float *__restrict__ part_l,
    float *__restrict__ part_acc, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ tile_q0,
    const int *__restrict__ tile_qn, const int *__restrict__ tile_seq,
    const int *__restrict__ tile_pos0, float scale, int Hq, int Hkv, int D,
    int BS, int MAXB, int GQ, int NS) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int split = blockIdx.z;
  const int hk = h / GQ;
  const int tid = threadIdx.x;
  const int q0 = tile_q0[tile];
  const int qn = tile_qn[tile];
  const int seq = tile_seq[tile];
  const int pos0 = tile_pos0[tile];
  const int kv_limit = pos0 + qn;
  const int chunks = (kv_limit + CHUNK - 1) / CHUNK;
  const int per = (chunks + NS - 1) / NS;
  const int t0 = split * per * CHUNK;
  const int t1 = min(kv_limit, (split + 1) * per * CHUNK);

  __shared__ float q_s[QT * 128];
  __shared__ float p_s[QT][CHUNK];
  __shared__ float scratch[8];

  for (int i = tid; i < qn * D; i += blockDim.x) {
    int qi = i / D, d = i % D;
    q_s[qi * D + d] = bf2f(q[((long)(q0 + qi) * Hq + h) * D + d]) * scale;
  }
  __syncthreads();

  float m[QT], l[QT], acc[QT];
#pragma unroll
  for (int qi = 0; qi < QT; ++qi) {
    m[qi] = -INFINITY;
    l[qi] = 0.f;
    acc[qi] = 0.f;
  }

  const long panel_stride = (long)Hkv * BS * D;
  for (int start = t0; start < t1; start += CHUNK) {
    const int clen = min(CHUNK, t1 - start);
    if (tid < clen) {
      const int token = start + tid;
      const long blk = bt[(long)seq * MAXB + token / BS];
      const bf16 *krow =
          kc + blk * panel_stride + ((long)hk * BS + token % BS) * D;
      float dots[QT];
#pragma unroll
      for (int qi = 0; qi < QT; ++qi) dots[qi] = 0.f;
      for (int d8 = 0; d8 < D / 8; ++d8) {
        uint4 kv = reinterpret_cast<const uint4 *>(krow)[d8];
        float kf[8];
        unpack_bf16x2(kv.x, kf[0], kf[1]);
        unpack_bf16x2(kv.y, kf[2], kf[3]);
        unpack_bf16x2(kv.z, kf[4], kf[5]);
        unpack_bf16x2(kv.w, kf[6], kf[7]);
#pragma unroll
        for (int qi = 0; qi < QT; ++qi) {
          if (qi >= qn) break;
          const float *qg = q_s + qi * D + d8 * 8;
#pragma unroll
          for (int k = 0; k < 8; ++k)
            dots[qi] = fmaf(kf[k], qg[k], dots[qi]);
        }
      }
#pragma unroll
      for (int qi = 0; qi < QT; ++qi) {
        if (qi >= qn) break;
        p_s[qi][tid] = (token <= pos0 + qi) ? dots[qi] : -INFINITY;
      }
    }
    __syncthreads();

#pragma unroll
    for (int qi = 0; qi < QT; ++qi) {
      if (qi >= qn) break;
      float mine = (tid < clen) ? p_s[qi][tid] : -INFINITY;
      float cmax = block_max(mine, scratch);
      if (cmax == -INFINITY) {
        if (tid < clen) p_s[qi][tid] = 0.f;
        __syncthreads();
        continue;
      }
      float mn = fmaxf(m[qi], cmax);
      float alpha = (m[qi] == -INFINITY) ? 0.f : __expf(m[qi] - mn);
      float p = (tid < clen && p_s[qi][tid] != -INFINITY)
                    ? __expf(p_s[qi][tid] - mn) : 0.f;
      if (tid < clen) p_s[qi][tid] = p;
      float psum = block_sum(p, scratch);
      l[qi] = l[qi] * alpha + psum;
      acc[qi] *= alpha;
      m[qi] = mn;
    }
    __syncthreads();

    if (tid < D) {
      for (int i = 0; i < clen; ++i) {
        const int token = start + i;
        const long blk = bt[(long)seq * MAXB + token / BS];
        const bf16 *vrow =
            vc + blk * panel_stride + ((long)hk * BS + token % BS) * D;
        const float v = bf2f(vrow[tid]);
#pragma unroll
        for (int qi = 0; qi < QT; ++qi) {
          if (qi >= qn) break;
          acc[qi] = fmaf(p_s[qi][i], v, acc[qi]);
        }
      }
    }
    __syncthreads();
  }

  const bool dead = (t0 >= kv_limit);
  if (tid < D) {
#pragma unroll
    for (int qi = 0; qi < QT; ++qi) {
      if (qi >= qn) break;
      const long base = (((long)tile * Hq + h) * NS + split) * QT + qi;
      if (tid == 0) {
        part_m[base] = dead ? -INFINITY : m[qi];
        part_l[base] = dead ? 0.f : l[qi];
      }
      part_acc[base * D + tid] = dead ? 0.f : acc[qi];
    }
  }
}

// Combine prefill split partials.  Grid: (ntiles, Hq, QT); block:
// DECODE_BLOCK.  One workgroup per query row — small grammar chunks
// launch few (tile, head) pairs, so the row loop must not serialize.
extern "C" __global__ void __launch_bounds__(DECODE_BLOCK)
paged_attn_prefill_reduce_kernel(bf16 *__restrict__ out,
                                 const float *__restrict__ part_m,
                                 const float *__restrict__ part_l,
                                 const float *__restrict__ part_acc,
                                 const int *__restrict__ tile_q0,
                                 const int *__restrict__ tile_qn,
                                 int Hq, int D, int NS) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int qi = blockIdx.z;
  const int tid = threadIdx.x;
  const int q0 = tile_q0[tile];
  if (qi >= tile_qn[tile]) return;
  __shared__ float scratch[8];

  const long base0 = (((long)tile * Hq + h) * NS) * QT + qi;
  float M = -INFINITY;
  for (int i = tid; i < NS; i += blockDim.x)
    M = fmaxf(M, part_m[base0 + (long)i * QT]);
  M = block_max(M, scratch);
  float L = 0.f;
  for (int i = tid; i < NS; i += blockDim.x) {
    float mi = part_m[base0 + (long)i * QT];
    L += (mi == -INFINITY) ? 0.f : part_l[base0 + (long)i * QT] * __expf(mi - M);
  }
  L = block_sum(L, scratch);
  if (tid < D) {
    float o = 0.f;
    for (int i = 0; i < NS; ++i) {
      float mi = part_m[base0 + (long)i * QT];
      if (mi == -INFINITY) continue;
      o = fmaf(part_acc[(base0 + (long)i * QT) * D + tid], __expf(mi - M), o);
    }
    out[((long)(q0 + qi) * Hq + h) * D + tid] = f2bf(L > 0.f ? o / L : 0.f);
  }
}

// ---------------------------------------------------------------------------
// EXPERIMENTAL (next-round validation): decode split v2 — 256-token chunks
// with 256 threads (half the barrier count per KV token), chunk block-ids
// precomputed once into LDS, K and V staged through LDS as in v1.
//   part layout identical to paged_attn_decode_split_kernel; reuses the
//   same reduce kernel.
// Grid: (B, Hkv, NS); block 256.  Requires DCHUNK % BS == 0.
// ---------------------------------------------------------------------------
#define DCHUNK2 256
#define DBLOCK2 256

extern "C" __global__ void __launch_bounds__(DBLOCK2)
paged_attn_decode_split2_kernel(
    float *__restrict__ part_m, float *__restrict__ part_l,
    float *__restrict__ part_acc, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ ctx, float scale,
    int Hq, int Hkv, int D, int BS, int MAXB, int GQ, int NS) {
  const int b = blockIdx.x;
  const int hk = blockIdx.y;
  const int split = blockIdx.z;
  const int tid = threadIdx.x;
  const int len = ctx[b];
  const int chunks = (len + DCHUNK2 - 1) / DCHUNK2;
  const int per = (chunks + NS - 1) / NS;
  const int t0 = split * per * DCHUNK2;
  const int t1 = min(len, (split + 1) * per * DCHUNK2);
  const bool dead = (t0 >= len);

  __shared__ float q_s[MAX_GQ * 128];
  __shared__ float p_s[MAX_GQ][DCHUNK2];
  __shared__ bf16 k_s[DCHUNK2 * KSTRIDE];
  __shared__ long blk_s[DCHUNK2 / 16 + 1];   // BS >= 16 -> <= 17 blocks
  __shared__ float scratch[8];

  for (int i = tid; i < GQ * D; i += blockDim.x) {
    int g = i / D, d = i % D;
    q_s[g * D + d] = bf2f(q[((long)b * Hq + hk * GQ + g) * D + d]) * scale;
  }

  float m[MAX_GQ], l[MAX_GQ], acc[MAX_GQ];
#pragma unroll
  for (int g = 0; g < MAX_GQ; ++g) {
    m[g] = -INFINITY;
    l[g] = 0.f;
    acc[g] = 0.f;
  }

  const long panel_stride = (long)Hkv * BS * D;
  const int dvecs = D / 8;
  const int blocks_per_chunk = DCHUNK2 / BS;
  for (int start = t0; start < t1; start += DCHUNK2) {
    const int clen = min(DCHUNK2, t1 - start);
    // chunk block ids once (start is DCHUNK2-aligned and DCHUNK2 % BS == 0)
    if (tid <= (clen - 1) / BS)
      blk_s[tid] = bt[(long)b * MAXB + start / BS + tid];
    __syncthreads();

    for (int i = tid; i < clen * dvecs; i += blockDim.x) {
      const int key = i / dvecs, d8 = i % dvecs;
      const long blk = blk_s[key / BS];
      const uint4 kv = reinterpret_cast<const uint4 *>(
          kc + blk * panel_stride + ((long)hk * BS + (start + key) % BS) * D)[d8];
      reinterpret_cast<uint4 *>(k_s + key * KSTRIDE + d8 * 8)[0] = kv;
    }
    __syncthreads();

    if (tid < clen) {
      float dots[MAX_GQ];
#pragma unroll
      for (int g = 0; g < MAX_GQ; ++g) dots[g] = 0.f;
      const bf16 *krow = k_s + tid * KSTRIDE;
      for (int d8 = 0; d8 < dvecs; ++d8) {
        uint4 kv = reinterpret_cast<const uint4 *>(krow)[d8];
        float kf[8];
        unpack_bf16x2(kv.x, kf[0], kf[1]);
        unpack_bf16x2(kv.y, kf[2], kf[3]);
        unpack_bf16x2(kv.z, kf[4], kf[5]);
        unpack_bf16x2(kv.w, kf[6], kf[7]);
#pragma unroll
        for (int g = 0; g < MAX_GQ; ++g) {
          if (g >= GQ) break;
          const float *qg = q_s + g * D + d8 * 8;
#pragma unroll
          for (int kx = 0; kx < 8; ++kx)
            dots[g] = fmaf(kf[kx], qg[kx], dots[g]);
        }
      }
#pragma unroll
      for (int g = 0; g < MAX_GQ; ++g) {
        if (g >= GQ) break;
        p_s[g][tid] = dots[g];
      }
    }
    __syncthreads();

    for (int g = 0; g < GQ; ++g) {
      float mine = (tid < clen) ? p_s[g][tid] : -INFINITY;
      float cmax = block_max(mine, scratch);
      float mn = fmaxf(m[g], cmax);
      float alpha = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - mn);
      float p = (tid < clen && mn != -INFINITY)
                    ? __expf(p_s[g][tid] - mn) : 0.f;
      if (tid < clen) p_s[g][tid] = p;
      float psum = block_sum(p, scratch);
      l[g] = l[g] * alpha + psum;
      acc[g] *= alpha;
      m[g] = mn;
    }
    __syncthreads();

    // V through LDS (time-shares k_s)
    for (int i = tid; i < clen * dvecs; i += blockDim.x) {
      const int key = i / dvecs, d8 = i % dvecs;
      const long blk = blk_s[key / BS];
      const uint4 vv = reinterpret_cast<const uint4 *>(
          vc + blk * panel_stride + ((long)hk * BS + (start + key) % BS) * D)[d8];
      reinterpret_cast<uint4 *>(k_s + key * KSTRIDE + d8 * 8)[0] = vv;
    }
    __syncthreads();

    if (tid < D) {
      for (int i = 0; i < clen; ++i) {
        const float v = bf2f(k_s[i * KSTRIDE + tid]);
#pragma unroll
        for (int g = 0; g < MAX_GQ; ++g) {
          if (g >= GQ) break;
          acc[g] = fmaf(p_s[g][i], v, acc[g]);
        }
      }
    }
    __syncthreads();
  }

  if (tid < D) {
#pragma unroll
    for (int g = 0; g < MAX_GQ; ++g) {
      if (g >= GQ) break;
      const int h = hk * GQ + g;
      const long base = ((long)b * Hq + h) * NS + split;
      if (tid == 0) {
        part_m[base] = dead ? -INFINITY : m[g];
        part_l[base] = dead ? 0.f : l[g];
      }
      part_acc[base * D + tid] = dead ? 0.f : acc[g];
    }
  }
}
