// MFMA-tiled paged prefill attention for gfx950 (CDNA4).
//
// The VALU prefill kernel profiles at 71% of bench GPU time
// (profiles/r01_bench_kernel_stats.csv): dot products on the vector ALU at
// ~7% efficiency.  This kernel moves QK^T and P·V onto the matrix cores
// with the guide's attention structure (cdna_hip_programming.md §3, §5.5):
//
//   * one workgroup (4 waves, 256 threads) per (16-query tile, q-head),
//     walking the KV context in 64-key chunks;
//   * K staged row-major and V staged TRANSPOSED in LDS with +8 element row
//     padding so every MFMA fragment read is bank-conflict-free;
//   * S = Q·K^T via v_mfma_f32_16x16x32_bf16 (each wave owns one 16-key
//     block, K-loop over D=128 in 4 steps);
//   * online softmax in LDS (persistent m/l per query row), P written back
//     as bf16 and re-read in the A-fragment layout;
//   * O += P·V accumulated in C fragments across chunks with per-chunk
//     alpha rescaling (textbook order: P of a chunk is exponentiated only
//     after the max decision that covers it — T13 hazard avoided).
//
// Fragment maps (cdna4_isa.md §10, 16x16x32 bf16):
//   A: row = lane&15,           k = (lane>>4)*8 + i   (8 contiguous)
//   B: col = lane&15,           k = (lane>>4)*8 + i
//   C/D: col = lane&15,         row = (lane>>4)*4 + reg
//
// Requires D == 128; host dispatch falls back to the VALU kernels
// otherwise (quoracle_amd/ops/dispatch.py).

#include "common.h"

#define MF_KCHUNK 64
#define MF_QT 16
#define MF_D 128
// LDS row pads keep fragment reads conflict-free (see dispatch notes)
#define KP (MF_D + 8)        // k_s / q_s row stride (elements)
#define VP (MF_KCHUNK + 8)   // vt_s / p_s row stride
#define SP (MF_KCHUNK + 4)   // s_s row stride (f32 words)
#define VR (MF_D + 6)        // row-major V stride (67 dwords, odd: spreads
                             // the PV key-gather across all LDS banks)

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4_t __attribute__((ext_vector_type(4)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

extern "C" __global__ void __launch_bounds__(256)
paged_attn_prefill_mfma_kernel(
    bf16 *__restrict__ out, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ tile_q0,
    const int *__restrict__ tile_qn, const int *__restrict__ tile_seq,
    const int *__restrict__ tile_pos0, float scale, int Hq, int Hkv, int BS,
    int MAXB, int GQ) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int hk = h / GQ;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int q0 = tile_q0[tile];
  const int qn = tile_qn[tile];
  const int seq = tile_seq[tile];
  const int pos0 = tile_pos0[tile];
  const int kv_limit = pos0 + qn;

  // q_s/k_s: XOR-swizzled [row][128] images (elem block ^ (row&15)<<3)
  __shared__ bf16 q_s[MF_QT * MF_D];
  __shared__ bf16 k_s[MF_KCHUNK * MF_D];
  __shared__ bf16 v_s[MF_KCHUNK * VR];   // row-major V (see VR note)
  __shared__ float s_s[MF_QT * SP];
  __shared__ bf16 p_s[MF_QT * VP];
  __shared__ float m_s[MF_QT], l_s[MF_QT], alpha_s[MF_QT];

  // ---- load the Q tile (rows >= qn zeroed) --------------------------------
  for (int i = tid; i < MF_QT * MF_D / 8; i += 256) {
    const int r = (i * 8) / MF_D, c = (i * 8) % MF_D;
    uint4 val = make_uint4(0, 0, 0, 0);
    if (r < qn)
      val = reinterpret_cast<const uint4 *>(
          q + ((long)(q0 + r) * Hq + h) * MF_D + c)[0];
    reinterpret_cast<uint4 *>(
        q_s + r * MF_D + (c ^ ((r & 15) << 3)))[0] = val;
  }
  if (tid < MF_QT) {
    m_s[tid] = -INFINITY;
    l_s[tid] = 0.f;
  }
  __syncthreads();

  // O accumulator fragments: wave w owns output cols [w*32, w*32+32)
  f32x4_t o_acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4_t o_acc1 = {0.f, 0.f, 0.f, 0.f};

  const long panel_stride = (long)Hkv * BS * MF_D;
  const int a_row = lane & 15;           // A/B col or row
  const int a_koff = (lane >> 4) * 8;    // k offset within 32-wide K slice
  const int c_col = lane & 15;
  const int c_row0 = (lane >> 4) * 4;

  // T14 software pipeline (guide §5.5): chunk c+1's K/V rows ride in
  // registers while chunk c computes; the LDS write happens after the
  // barrier, so HBM latency hides under the MFMA phases.
  uint4 kreg[4], vreg[4];
  auto issue_loads = [&](int start_, int limit_) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = tid + it * 256;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      uint4 kv = make_uint4(0, 0, 0, 0), vv = make_uint4(0, 0, 0, 0);
      const int token = start_ + key;
      if (token < limit_) {
        const long blk = bt[(long)seq * MAXB + token / BS];
        const long off =
            blk * panel_stride + ((long)hk * BS + token % BS) * MF_D + d;
        kv = reinterpret_cast<const uint4 *>(kc + off)[0];
        vv = reinterpret_cast<const uint4 *>(vc + off)[0];
      }
      kreg[it] = kv;
      vreg[it] = vv;
    }
  };
  auto write_staged = [&]() {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = tid + it * 256;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      reinterpret_cast<uint4 *>(
          k_s + key * MF_D + (d ^ ((key & 15) << 3)))[0] = kreg[it];
      const uint *vw = reinterpret_cast<const uint *>(&vreg[it]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        reinterpret_cast<uint *>(v_s + key * VR + d)[j] = vw[j];
    }
  };

  issue_loads(0, kv_limit);
  for (int start = 0; start < kv_limit; start += MF_KCHUNK) {
    const int clen = min(MF_KCHUNK, kv_limit - start);

    // staged K/V (loaded last iteration) -> LDS; prefetch next chunk
    write_staged();
    __syncthreads();
    if (start + MF_KCHUNK < kv_limit)
      issue_loads(start + MF_KCHUNK, kv_limit);

    // ---- S = Q·K^T (wave w: key block w*16..w*16+15) ----------------------
    {
      f32x4_t s_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < MF_D / 32; ++kk) {
        bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
            q_s + a_row * MF_D + ((kk * 32 + a_koff) ^ ((a_row & 15) << 3)));
        bf16x8_t b = *reinterpret_cast<const bf16x8_t *>(
            k_s + (wave * 16 + a_row) * MF_D
                + ((kk * 32 + a_koff) ^ ((a_row & 15) << 3)));
        s_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, s_acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = c_row0 + r;
        const int col = wave * 16 + c_col;
        const int token = start + col;
        const bool ok = (token <= pos0 + min(row, qn - 1)) && (col < clen);
        s_s[row * SP + col] = ok ? s_acc[r] * scale : -INFINITY;
      }
    }
    __syncthreads();

    // ---- online softmax (16 threads per query row) ------------------------
    {
      const int row = tid >> 4;
      const int sub = tid & 15;
      float v0 = s_s[row * SP + sub];
      float v1 = s_s[row * SP + sub + 16];
      float v2 = s_s[row * SP + sub + 32];
      float v3 = s_s[row * SP + sub + 48];
      float mymax = fmaxf(fmaxf(v0, v1), fmaxf(v2, v3));
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1)
        mymax = fmaxf(mymax, __shfl_xor(mymax, w, 16));
      const float m_old = m_s[row];
      const float mn = fmaxf(m_old, mymax);
      const float alpha = (m_old == -INFINITY) ? 0.f : __expf(m_old - mn);
      float p0 = (v0 == -INFINITY) ? 0.f : __expf(v0 - mn);
      float p1 = (v1 == -INFINITY) ? 0.f : __expf(v1 - mn);
      float p2 = (v2 == -INFINITY) ? 0.f : __expf(v2 - mn);
      float p3 = (v3 == -INFINITY) ? 0.f : __expf(v3 - mn);
      float psum = p0 + p1 + p2 + p3;
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1)
        psum += __shfl_xor(psum, w, 16);
      if (sub == 0) {
        l_s[row] = l_s[row] * alpha + psum;
        m_s[row] = mn;
        alpha_s[row] = alpha;
      }
      p_s[row * VP + sub] = f2bf(p0);
      p_s[row * VP + sub + 16] = f2bf(p1);
      p_s[row * VP + sub + 32] = f2bf(p2);
      p_s[row * VP + sub + 48] = f2bf(p3);
    }
    __syncthreads();

    // ---- O rescale + O += P·V --------------------------------------------
    {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float a = alpha_s[c_row0 + r];
        o_acc0[r] *= a;
        o_acc1[r] *= a;
      }
#pragma unroll
      for (int kk = 0; kk < MF_KCHUNK / 32; ++kk) {
        bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
            p_s + a_row * VP + kk * 32 + a_koff);
        bf16x8_t b0, b1;
#pragma unroll
        for (int t = 0; t < 8; ++t) {
          const bf16 *vrow = v_s + (kk * 32 + a_koff + t) * VR + wave * 32;
          b0[t] = vrow[c_col];
          b1[t] = vrow[16 + c_col];
        }
        o_acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b0, o_acc0, 0, 0, 0);
        o_acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b1, o_acc1, 0, 0, 0);
      }
    }
    __syncthreads();   // next chunk restages k_s/vt_s
  }

  // ---- epilogue -----------------------------------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = c_row0 + r;
    if (row >= qn) continue;
    const float denom = l_s[row] > 0.f ? l_s[row] : 1.f;
    out[((long)(q0 + row) * Hq + h) * MF_D + wave * 32 + c_col] =
        f2bf(o_acc0[r] / denom);
    out[((long)(q0 + row) * Hq + h) * MF_D + wave * 32 + 16 + c_col] =
        f2bf(o_acc1[r] / denom);
  }
}

// ---------------------------------------------------------------------------
// Context-split MFMA prefill: small chunks (grammar forced-runs) have only
// a handful of tiles, so (ntiles, Hq) leaves the chip empty while each WG
// walks the whole cached context.  Partition the walk over NS workgroups;
// partials land in the SAME layout as the VALU split kernel
// (part_m/l: [ntiles, Hq, NS, QT], part_acc: [..., D]) so the existing
// paged_attn_prefill_reduce_kernel combines them.
// Grid: (ntiles, Hq, NS); block 256 (4 waves).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
paged_attn_prefill_mfma_split_kernel(
    float *__restrict__ part_m, float *__restrict__ part_l,
    float *__restrict__ part_acc, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ tile_q0,
    const int *__restrict__ tile_qn, const int *__restrict__ tile_seq,
    const int *__restrict__ tile_pos0, float scale, int Hq, int Hkv, int BS,
    int MAXB, int GQ, int NS) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int split = blockIdx.z;
  const int hk = h / GQ;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int q0 = tile_q0[tile];
  const int qn = tile_qn[tile];
  const int seq = tile_seq[tile];
  const int pos0 = tile_pos0[tile];
  const int kv_limit = pos0 + qn;
  const int chunks = (kv_limit + MF_KCHUNK - 1) / MF_KCHUNK;
  const int per = (chunks + NS - 1) / NS;
  const int c0 = split * per * MF_KCHUNK;
  const int c1 = min(kv_limit, (split + 1) * per * MF_KCHUNK);
  const bool dead = (c0 >= kv_limit);

  // q_s/k_s: XOR-swizzled [row][128] images (elem block ^ (row&15)<<3)
  __shared__ bf16 q_s[MF_QT * MF_D];
  __shared__ bf16 k_s[MF_KCHUNK * MF_D];
  __shared__ bf16 v_s[MF_KCHUNK * VR];   // row-major V (see VR note)
  __shared__ float s_s[MF_QT * SP];
  __shared__ bf16 p_s[MF_QT * VP];
  __shared__ float m_s[MF_QT], l_s[MF_QT], alpha_s[MF_QT];

  for (int i = tid; i < MF_QT * MF_D / 8; i += 256) {
    const int r = (i * 8) / MF_D, c = (i * 8) % MF_D;
    uint4 val = make_uint4(0, 0, 0, 0);
    if (r < qn)
      val = reinterpret_cast<const uint4 *>(
          q + ((long)(q0 + r) * Hq + h) * MF_D + c)[0];
    reinterpret_cast<uint4 *>(
        q_s + r * MF_D + (c ^ ((r & 15) << 3)))[0] = val;
  }
  if (tid < MF_QT) {
    m_s[tid] = -INFINITY;
    l_s[tid] = 0.f;
  }
  __syncthreads();

  f32x4_t o_acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4_t o_acc1 = {0.f, 0.f, 0.f, 0.f};

  const long panel_stride = (long)Hkv * BS * MF_D;
  const int a_row = lane & 15;
  const int a_koff = (lane >> 4) * 8;
  const int c_col = lane & 15;
  const int c_row0 = (lane >> 4) * 4;

  // T14 software pipeline (guide §5.5): chunk c+1's K/V rows ride in
  // registers while chunk c computes; the LDS write happens after the
  // barrier, so HBM latency hides under the MFMA phases.
  uint4 kreg[4], vreg[4];
  auto issue_loads = [&](int start_, int limit_) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = tid + it * 256;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      uint4 kv = make_uint4(0, 0, 0, 0), vv = make_uint4(0, 0, 0, 0);
      const int token = start_ + key;
      if (token < limit_) {
        const long blk = bt[(long)seq * MAXB + token / BS];
        const long off =
            blk * panel_stride + ((long)hk * BS + token % BS) * MF_D + d;
        kv = reinterpret_cast<const uint4 *>(kc + off)[0];
        vv = reinterpret_cast<const uint4 *>(vc + off)[0];
      }
      kreg[it] = kv;
      vreg[it] = vv;
    }
  };
  auto write_staged = [&]() {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = tid + it * 256;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      reinterpret_cast<uint4 *>(
          k_s + key * MF_D + (d ^ ((key & 15) << 3)))[0] = kreg[it];
      const uint *vw = reinterpret_cast<const uint *>(&vreg[it]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        reinterpret_cast<uint *>(v_s + key * VR + d)[j] = vw[j];
    }
  };

  issue_loads(c0, c1);
  for (int start = c0; start < c1; start += MF_KCHUNK) {
    const int clen = min(MF_KCHUNK, c1 - start);

    write_staged();
    __syncthreads();
    if (start + MF_KCHUNK < c1)
      issue_loads(start + MF_KCHUNK, c1);

    {
      f32x4_t s_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < MF_D / 32; ++kk) {
        bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
            q_s + a_row * MF_D + ((kk * 32 + a_koff) ^ ((a_row & 15) << 3)));
        bf16x8_t b = *reinterpret_cast<const bf16x8_t *>(
            k_s + (wave * 16 + a_row) * MF_D
                + ((kk * 32 + a_koff) ^ ((a_row & 15) << 3)));
        s_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, s_acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = c_row0 + r;
        const int col = wave * 16 + c_col;
        const int token = start + col;
        const bool ok = (token <= pos0 + min(row, qn - 1)) && (col < clen);
        s_s[row * SP + col] = ok ? s_acc[r] * scale : -INFINITY;
      }
    }
    __syncthreads();

    {
      const int row = tid >> 4;
      const int sub = tid & 15;
      float v0 = s_s[row * SP + sub];
      float v1 = s_s[row * SP + sub + 16];
      float v2 = s_s[row * SP + sub + 32];
      float v3 = s_s[row * SP + sub + 48];
      float mymax = fmaxf(fmaxf(v0, v1), fmaxf(v2, v3));
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1)
        mymax = fmaxf(mymax, __shfl_xor(mymax, w, 16));
      const float m_old = m_s[row];
      const float mn = fmaxf(m_old, mymax);
      const float alpha = (m_old == -INFINITY) ? 0.f : __expf(m_old - mn);
      float p0 = (v0 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v0 - mn);
      float p1 = (v1 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v1 - mn);
      float p2 = (v2 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v2 - mn);
      float p3 = (v3 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v3 - mn);
      float psum = p0 + p1 + p2 + p3;
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1)
        psum += __shfl_xor(psum, w, 16);
      if (sub == 0) {
        l_s[row] = l_s[row] * alpha + psum;
        m_s[row] = mn;
        alpha_s[row] = alpha;
      }
      p_s[row * VP + sub] = f2bf(p0);
      p_s[row * VP + sub + 16] = f2bf(p1);
      p_s[row * VP + sub + 32] = f2bf(p2);
      p_s[row * VP + sub + 48] = f2bf(p3);
    }
    __syncthreads();

    {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float a = alpha_s[c_row0 + r];
        o_acc0[r] *= a;
        o_acc1[r] *= a;
      }
#pragma unroll
      for (int kk = 0; kk < MF_KCHUNK / 32; ++kk) {
        bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
            p_s + a_row * VP + kk * 32 + a_koff);
        bf16x8_t b0, b1;
#pragma unroll
        for (int t = 0; t < 8; ++t) {
          const bf16 *vrow = v_s + (kk * 32 + a_koff + t) * VR + wave * 32;
          b0[t] = vrow[c_col];
          b1[t] = vrow[16 + c_col];
        }
        o_acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b0, o_acc0, 0, 0, 0);
        o_acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b1, o_acc1, 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // partials out (unnormalized; reduce kernel combines across splits)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = c_row0 + r;
    if (row >= MF_QT) continue;
    const long base = (((long)tile * Hq + h) * NS + split) * MF_QT + row;
    if (c_col == 0 && wave == 0) {
      part_m[base] = dead ? -INFINITY : m_s[row];
      part_l[base] = dead ? 0.f : l_s[row];
    }
    part_acc[base * MF_D + wave * 32 + c_col] = dead ? 0.f : o_acc0[r];
    part_acc[base * MF_D + wave * 32 + 16 + c_col] = dead ? 0.f : o_acc1[r];
  }
}

// ---------------------------------------------------------------------------
// 32-row Q-tile variant: 8 waves (512 threads) per WG cover 32 query rows,
// so each staged 64-key K/V chunk serves 2x the query work — half the LDS
// staging traffic and barrier count per query token vs the 16-row kernel.
// Wave w: S-phase tile (qblock=w>>2, keyblock=w&3); PV-phase d-columns
// (w&3)*32..+31 for q rows (w>>2)*16..+15.
// Grid: (ntiles32, Hq); block 512.  Tile metadata uses 32-row tiles
// (tile32_* built by the engine alongside the 16-row set).
// ---------------------------------------------------------------------------
#define MF2_QT 32

extern "C" __global__ void __launch_bounds__(512)
paged_attn_prefill_mfma32_kernel(
    bf16 *__restrict__ out, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ tile_q0,
    const int *__restrict__ tile_qn, const int *__restrict__ tile_seq,
    const int *__restrict__ tile_pos0, float scale, int Hq, int Hkv, int BS,
    int MAXB, int GQ) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int hk = h / GQ;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int q0 = tile_q0[tile];
  const int qn = tile_qn[tile];
  const int seq = tile_seq[tile];
  const int pos0 = tile_pos0[tile];
  const int kv_limit = pos0 + qn;

  // q_s/k_s: XOR-swizzled [row][128] images (elem block ^ (row&15)<<3)
  __shared__ bf16 q_s[MF2_QT * MF_D];
  __shared__ bf16 k_s[MF_KCHUNK * MF_D];
  __shared__ bf16 v_s[MF_KCHUNK * VR];   // row-major V (see VR note)
  __shared__ float s_s[MF2_QT * SP];
  __shared__ bf16 p_s[MF2_QT * VP];
  __shared__ float m_s[MF2_QT], l_s[MF2_QT], alpha_s[MF2_QT];

  for (int i = tid; i < MF2_QT * MF_D / 8; i += 512) {
    const int r = (i * 8) / MF_D, c = (i * 8) % MF_D;
    uint4 val = make_uint4(0, 0, 0, 0);
    if (r < qn)
      val = reinterpret_cast<const uint4 *>(
          q + ((long)(q0 + r) * Hq + h) * MF_D + c)[0];
    reinterpret_cast<uint4 *>(
        q_s + r * MF_D + (c ^ ((r & 15) << 3)))[0] = val;
  }
  if (tid < MF2_QT) {
    m_s[tid] = -INFINITY;
    l_s[tid] = 0.f;
  }
  __syncthreads();

  f32x4_t o_acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4_t o_acc1 = {0.f, 0.f, 0.f, 0.f};

  const long panel_stride = (long)Hkv * BS * MF_D;
  const int a_row = lane & 15;
  const int a_koff = (lane >> 4) * 8;
  const int c_col = lane & 15;
  const int c_row0 = (lane >> 4) * 4;
  const int qblock = wave >> 2;          // 0..1: 16-row half owned in S & PV
  const int kblock = wave & 3;           // S-phase key block / PV d-block

  // T14 pipeline at 2 iterations per thread (512 threads)
  uint4 kreg[2], vreg[2];
  auto issue_loads = [&](int start_, int limit_) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int i = tid + it * 512;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      uint4 kv = make_uint4(0, 0, 0, 0), vv = make_uint4(0, 0, 0, 0);
      const int token = start_ + key;
      if (token < limit_) {
        const long blk = bt[(long)seq * MAXB + token / BS];
        const long off =
            blk * panel_stride + ((long)hk * BS + token % BS) * MF_D + d;
        kv = reinterpret_cast<const uint4 *>(kc + off)[0];
        vv = reinterpret_cast<const uint4 *>(vc + off)[0];
      }
      kreg[it] = kv;
      vreg[it] = vv;
    }
  };
  auto write_staged = [&]() {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int i = tid + it * 512;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      reinterpret_cast<uint4 *>(
          k_s + key * MF_D + (d ^ ((key & 15) << 3)))[0] = kreg[it];
      const uint *vw = reinterpret_cast<const uint *>(&vreg[it]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        reinterpret_cast<uint *>(v_s + key * VR + d)[j] = vw[j];
    }
  };

  issue_loads(0, kv_limit);
  for (int start = 0; start < kv_limit; start += MF_KCHUNK) {
    const int clen = min(MF_KCHUNK, kv_limit - start);
    write_staged();
    __syncthreads();
    if (start + MF_KCHUNK < kv_limit)
      issue_loads(start + MF_KCHUNK, kv_limit);

    // ---- S = Q·K^T: wave handles (qblock, kblock) ------------------------
    {
      f32x4_t s_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < MF_D / 32; ++kk) {
        bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
            q_s + (qblock * 16 + a_row) * MF_D
                + ((kk * 32 + a_koff) ^ ((a_row & 15) << 3)));
        bf16x8_t b = *reinterpret_cast<const bf16x8_t *>(
            k_s + (kblock * 16 + a_row) * MF_D
                + ((kk * 32 + a_koff) ^ ((a_row & 15) << 3)));
        s_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, s_acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qblock * 16 + c_row0 + r;
        const int col = kblock * 16 + c_col;
        const int token = start + col;
        const bool ok = (token <= pos0 + min(row, qn - 1)) && (col < clen);
        s_s[row * SP + col] = ok ? s_acc[r] * scale : -INFINITY;
      }
    }
    __syncthreads();

    // ---- online softmax (16 threads per row, 32 rows) --------------------
    {
      const int row = tid >> 4;
      const int sub = tid & 15;
      float v0 = s_s[row * SP + sub];
      float v1 = s_s[row * SP + sub + 16];
      float v2 = s_s[row * SP + sub + 32];
      float v3 = s_s[row * SP + sub + 48];
      float mymax = fmaxf(fmaxf(v0, v1), fmaxf(v2, v3));
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1)
        mymax = fmaxf(mymax, __shfl_xor(mymax, w, 16));
      const float m_old = m_s[row];
      const float mn = fmaxf(m_old, mymax);
      const float alpha = (m_old == -INFINITY) ? 0.f : __expf(m_old - mn);
      float p0 = (v0 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v0 - mn);
      float p1 = (v1 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v1 - mn);
      float p2 = (v2 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v2 - mn);
      float p3 = (v3 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v3 - mn);
      float psum = p0 + p1 + p2 + p3;
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1)
        psum += __shfl_xor(psum, w, 16);
      if (sub == 0) {
        l_s[row] = l_s[row] * alpha + psum;
        m_s[row] = mn;
        alpha_s[row] = alpha;
      }
      p_s[row * VP + sub] = f2bf(p0);
      p_s[row * VP + sub + 16] = f2bf(p1);
      p_s[row * VP + sub + 32] = f2bf(p2);
      p_s[row * VP + sub + 48] = f2bf(p3);
    }
    __syncthreads();

    // ---- O rescale + O += P·V: wave covers d cols kblock*32..+31 ---------
    {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float a = alpha_s[qblock * 16 + c_row0 + r];
        o_acc0[r] *= a;
        o_acc1[r] *= a;
      }
#pragma unroll
      for (int kk = 0; kk < MF_KCHUNK / 32; ++kk) {
        bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
            p_s + (qblock * 16 + a_row) * VP + kk * 32 + a_koff);
        bf16x8_t b0, b1;
#pragma unroll
        for (int t = 0; t < 8; ++t) {
          const bf16 *vrow = v_s + (kk * 32 + a_koff + t) * VR + kblock * 32;
          b0[t] = vrow[c_col];
          b1[t] = vrow[16 + c_col];
        }
        o_acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b0, o_acc0, 0, 0, 0);
        o_acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b1, o_acc1, 0, 0, 0);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = qblock * 16 + c_row0 + r;
    if (row >= qn) continue;
    const float denom = l_s[row] > 0.f ? l_s[row] : 1.f;
    out[((long)(q0 + row) * Hq + h) * MF_D + kblock * 32 + c_col] =
        f2bf(o_acc0[r] / denom);
    out[((long)(q0 + row) * Hq + h) * MF_D + kblock * 32 + 16 + c_col] =
        f2bf(o_acc1[r] / denom);
  }
}

// ---------------------------------------------------------------------------
// EXPERIMENTAL (next-round validation): 64-row Q tiles, 8 waves.  4x K/V
// staging reuse per query row vs the 16-row kernel.  Wave w: S-phase
// keyblock w&3 for qblocks {2*(w>>2), 2*(w>>2)+1}; PV-phase d-columns
// (w&3)*32..+31 for the same two qblocks.
// Grid: (ntiles64, Hq); block 512.
// ---------------------------------------------------------------------------
#define MF4_QT 64

extern "C" __global__ void __launch_bounds__(512)
paged_attn_prefill_mfma64_kernel(
    bf16 *__restrict__ out, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ tile_q0,
    const int *__restrict__ tile_qn, const int *__restrict__ tile_seq,
    const int *__restrict__ tile_pos0, float scale, int Hq, int Hkv, int BS,
    int MAXB, int GQ) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int hk = h / GQ;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int q0 = tile_q0[tile];
  const int qn = tile_qn[tile];
  const int seq = tile_seq[tile];
  const int pos0 = tile_pos0[tile];
  const int kv_limit = pos0 + qn;

  // V stays ROW-major in LDS (key-major, stride VR elems).  The old
  // transposed image vt_s[d][key] made the staging scatter a 32-way bank
  // conflict (8-row lane stride x 136-elem rows = bank step 32 mod 64);
  // row-major staging writes are conflict-free b32s, and VR = 134
  // (67 dwords, odd) spreads the PV B-fragment's scalar key-gather over
  // all banks: key groups land at +24k mod 64, d-columns at +c/2.
  // q_s/k_s: XOR-swizzled [row][128] images (guide G4: elem block index
  // ^ (row&15)<<3) — zero-conflict b128 A/B-fragment reads at the full
  // 64-lane phase, where +8 padding still left 2-way phase conflicts
  __shared__ bf16 q_s[MF4_QT * MF_D];
  __shared__ bf16 k_s[MF_KCHUNK * MF_D];
  __shared__ bf16 v_s[MF_KCHUNK * VR];
  __shared__ float s_s[MF4_QT * SP];
  __shared__ bf16 p_s[MF4_QT * VP];
  __shared__ float m_s[MF4_QT], l_s[MF4_QT], alpha_s[MF4_QT];

  for (int i = tid; i < MF4_QT * MF_D / 8; i += 512) {
    const int r = (i * 8) / MF_D, c = (i * 8) % MF_D;
    uint4 val = make_uint4(0, 0, 0, 0);
    if (r < qn)
      val = reinterpret_cast<const uint4 *>(
          q + ((long)(q0 + r) * Hq + h) * MF_D + c)[0];
    reinterpret_cast<uint4 *>(
        q_s + r * MF_D + (c ^ ((r & 15) << 3)))[0] = val;
  }
  if (tid < MF4_QT) {
    m_s[tid] = -INFINITY;
    l_s[tid] = 0.f;
  }
  __syncthreads();

  // two qblocks per wave: frags xN for qblock qb0 and qb0+1
  f32x4_t s_acc0, s_acc1;
  f32x4_t o_acc00 = {0.f, 0.f, 0.f, 0.f};   // qb0,   dcol kblock*32
  f32x4_t o_acc01 = {0.f, 0.f, 0.f, 0.f};   // qb0,   dcol kblock*32+16
  f32x4_t o_acc10 = {0.f, 0.f, 0.f, 0.f};   // qb0+1, dcol kblock*32
  f32x4_t o_acc11 = {0.f, 0.f, 0.f, 0.f};   // qb0+1, dcol kblock*32+16

  const long panel_stride = (long)Hkv * BS * MF_D;
  const int a_row = lane & 15;
  const int a_koff = (lane >> 4) * 8;
  const int c_col = lane & 15;
  const int c_row0 = (lane >> 4) * 4;
  const int kblock = wave & 3;
  const int qb0 = (wave >> 2) * 2;

  uint4 kreg[2], vreg[2];
  auto issue_loads = [&](int start_, int limit_) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int i = tid + it * 512;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      uint4 kv = make_uint4(0, 0, 0, 0), vv = make_uint4(0, 0, 0, 0);
      const int token = start_ + key;
      if (token < limit_) {
        const long blk = bt[(long)seq * MAXB + token / BS];
        const long off =
            blk * panel_stride + ((long)hk * BS + token % BS) * MF_D + d;
        kv = reinterpret_cast<const uint4 *>(kc + off)[0];
        vv = reinterpret_cast<const uint4 *>(vc + off)[0];
      }
      kreg[it] = kv;
      vreg[it] = vv;
    }
  };
  auto write_staged = [&]() {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int i = tid + it * 512;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      reinterpret_cast<uint4 *>(
          k_s + key * MF_D + (d ^ ((key & 15) << 3)))[0] = kreg[it];
      // V row-major: 4 b32 stores (VR row stride is 4-byte aligned only);
      // conflict-free — lanes sharing a key span banks 4 apart, the four
      // keys per instruction land at distinct bank residues (VR/2 odd)
      const uint *vw = reinterpret_cast<const uint *>(&vreg[it]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        reinterpret_cast<uint *>(v_s + key * VR + d)[j] = vw[j];
    }
  };

  issue_loads(0, kv_limit);
  for (int start = 0; start < kv_limit; start += MF_KCHUNK) {
    const int clen = min(MF_KCHUNK, kv_limit - start);
    write_staged();
    __syncthreads();
    if (start + MF_KCHUNK < kv_limit)
      issue_loads(start + MF_KCHUNK, kv_limit);

    {
      s_acc0 = (f32x4_t){0.f, 0.f, 0.f, 0.f};
      s_acc1 = (f32x4_t){0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < MF_D / 32; ++kk) {
        const int koff = (kk * 32 + a_koff) ^ ((a_row & 15) << 3);
        bf16x8_t b = *reinterpret_cast<const bf16x8_t *>(
            k_s + (kblock * 16 + a_row) * MF_D + koff);
        bf16x8_t a0 = *reinterpret_cast<const bf16x8_t *>(
            q_s + ((qb0 + 0) * 16 + a_row) * MF_D + koff);
        bf16x8_t a1 = *reinterpret_cast<const bf16x8_t *>(
            q_s + ((qb0 + 1) * 16 + a_row) * MF_D + koff);
        s_acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b, s_acc0, 0, 0, 0);
        s_acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b, s_acc1, 0, 0, 0);
      }
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const f32x4_t &sa = half ? s_acc1 : s_acc0;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = (qb0 + half) * 16 + c_row0 + r;
          const int col = kblock * 16 + c_col;
          const int token = start + col;
          const bool ok = (token <= pos0 + min(row, qn - 1)) && (col < clen);
          s_s[row * SP + col] = ok ? sa[r] * scale : -INFINITY;
        }
      }
    }
    __syncthreads();

    // softmax: 8 threads per row, 64 rows
    {
      const int row = tid >> 3;
      const int sub = tid & 7;
      float v[8];
      float mymax = -INFINITY;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        v[j] = s_s[row * SP + sub + j * 8];
        mymax = fmaxf(mymax, v[j]);
      }
#pragma unroll
      for (int w = 4; w >= 1; w >>= 1)
        mymax = fmaxf(mymax, __shfl_xor(mymax, w, 8));
      const float m_old = m_s[row];
      const float mn = fmaxf(m_old, mymax);
      const float alpha = (m_old == -INFINITY) ? 0.f : __expf(m_old - mn);
      float psum = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float p = (v[j] == -INFINITY || mn == -INFINITY)
                            ? 0.f : __expf(v[j] - mn);
        p_s[row * VP + sub + j * 8] = f2bf(p);
        psum += p;
      }
#pragma unroll
      for (int w = 4; w >= 1; w >>= 1)
        psum += __shfl_xor(psum, w, 8);
      if (sub == 0) {
        l_s[row] = l_s[row] * alpha + psum;
        m_s[row] = mn;
        alpha_s[row] = alpha;
      }
    }
    __syncthreads();

    {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float a0 = alpha_s[(qb0 + 0) * 16 + c_row0 + r];
        const float a1 = alpha_s[(qb0 + 1) * 16 + c_row0 + r];
        o_acc00[r] *= a0;
        o_acc01[r] *= a0;
        o_acc10[r] *= a1;
        o_acc11[r] *= a1;
      }
#pragma unroll
      for (int kk = 0; kk < MF_KCHUNK / 32; ++kk) {
        // B fragments gathered from row-major V: lane t-loop walks 8 keys
        // at a fixed d column (conflict-free by VR construction)
        bf16x8_t b0, b1;
#pragma unroll
        for (int t = 0; t < 8; ++t) {
          const bf16 *vrow = v_s + (kk * 32 + a_koff + t) * VR + kblock * 32;
          b0[t] = vrow[c_col];
          b1[t] = vrow[16 + c_col];
        }
        bf16x8_t a0 = *reinterpret_cast<const bf16x8_t *>(
            p_s + ((qb0 + 0) * 16 + a_row) * VP + kk * 32 + a_koff);
        bf16x8_t a1 = *reinterpret_cast<const bf16x8_t *>(
            p_s + ((qb0 + 1) * 16 + a_row) * VP + kk * 32 + a_koff);
        o_acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, o_acc00, 0, 0, 0);
        o_acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, o_acc01, 0, 0, 0);
        o_acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, o_acc10, 0, 0, 0);
        o_acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, o_acc11, 0, 0, 0);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int half = 0; half < 2; ++half) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (qb0 + half) * 16 + c_row0 + r;
      if (row >= qn) continue;
      const float denom = l_s[row] > 0.f ? l_s[row] : 1.f;
      const f32x4_t &oa = half ? o_acc10 : o_acc00;
      const f32x4_t &ob = half ? o_acc11 : o_acc01;
      out[((long)(q0 + row) * Hq + h) * MF_D + kblock * 32 + c_col] =
          f2bf(oa[r] / denom);
      out[((long)(q0 + row) * Hq + h) * MF_D + kblock * 32 + 16 + c_col] =
          f2bf(ob[r] / denom);
    }
  }
}

// ---------------------------------------------------------------------------
// MFMA flash-decode: the VALU split decode kernel is latency-bound (128
// serial P·V iterations per chunk, 2-wave workgroups) and measured 0.78 TB/s
// of the 8 TB/s HBM roofline.  This kernel reuses the MFMA prefill-split
// structure for decode: the GQ query heads sharing one KV head form the
// first GQ rows of a 16-row Q tile (rest zero-padded), K/V stream through
// LDS with the T14 issue-early/write-late pipeline, and both QK^T and P·V
// run on the matrix cores — no serial V walk, 4 waves per workgroup.
//   part_m/part_l: [B, Hq, NS] f32;  part_acc: [B, Hq, NS, D] f32
//   (identical to paged_attn_decode_split_kernel; combined by
//    paged_attn_decode_reduce_kernel).
// Grid: (B, Hkv, NS); block 256.  Requires D == 128, GQ <= 16.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
paged_attn_decode_mfma_kernel(
    float *__restrict__ part_m, float *__restrict__ part_l,
    float *__restrict__ part_acc, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ ctx, float scale,
    int Hq, int Hkv, int BS, int MAXB, int GQ, int NS) {
  const int b = blockIdx.x;
  const int hk = blockIdx.y;
  const int split = blockIdx.z;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int len = ctx[b];
  const int chunks = (len + MF_KCHUNK - 1) / MF_KCHUNK;
  const int per = (chunks + NS - 1) / NS;
  const int t0 = split * per * MF_KCHUNK;
  const int t1 = min(len, (split + 1) * per * MF_KCHUNK);
  const bool dead = (t0 >= len);

  // q_s/k_s: XOR-swizzled [row][128] images (elem block ^ (row&15)<<3)
  __shared__ bf16 q_s[MF_QT * MF_D];
  __shared__ bf16 k_s[MF_KCHUNK * MF_D];
  __shared__ bf16 v_s[MF_KCHUNK * VR];   // row-major V (see VR note)
  __shared__ float s_s[MF_QT * SP];
  __shared__ bf16 p_s[MF_QT * VP];
  __shared__ float m_s[MF_QT], l_s[MF_QT], alpha_s[MF_QT];

  // Q tile: rows 0..GQ-1 are this KV head's query heads, rest zero
  for (int i = tid; i < MF_QT * MF_D / 8; i += 256) {
    const int r = (i * 8) / MF_D, c = (i * 8) % MF_D;
    uint4 val = make_uint4(0, 0, 0, 0);
    if (r < GQ)
      val = reinterpret_cast<const uint4 *>(
          q + ((long)b * Hq + hk * GQ + r) * MF_D + c)[0];
    reinterpret_cast<uint4 *>(
        q_s + r * MF_D + (c ^ ((r & 15) << 3)))[0] = val;
  }
  if (tid < MF_QT) {
    m_s[tid] = -INFINITY;
    l_s[tid] = 0.f;
  }
  __syncthreads();

  f32x4_t o_acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4_t o_acc1 = {0.f, 0.f, 0.f, 0.f};

  const long panel_stride = (long)Hkv * BS * MF_D;
  const int a_row = lane & 15;
  const int a_koff = (lane >> 4) * 8;
  const int c_col = lane & 15;
  const int c_row0 = (lane >> 4) * 4;

  uint4 kreg[4], vreg[4];
  auto issue_loads = [&](int start_, int limit_) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = tid + it * 256;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      uint4 kv = make_uint4(0, 0, 0, 0), vv = make_uint4(0, 0, 0, 0);
      const int token = start_ + key;
      if (token < limit_) {
        const long blk = bt[(long)b * MAXB + token / BS];
        const long off =
            blk * panel_stride + ((long)hk * BS + token % BS) * MF_D + d;
        kv = reinterpret_cast<const uint4 *>(kc + off)[0];
        vv = reinterpret_cast<const uint4 *>(vc + off)[0];
      }
      kreg[it] = kv;
      vreg[it] = vv;
    }
  };
  auto write_staged = [&]() {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = tid + it * 256;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      reinterpret_cast<uint4 *>(
          k_s + key * MF_D + (d ^ ((key & 15) << 3)))[0] = kreg[it];
      const uint *vw = reinterpret_cast<const uint *>(&vreg[it]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        reinterpret_cast<uint *>(v_s + key * VR + d)[j] = vw[j];
    }
  };

  issue_loads(t0, t1);
  for (int start = t0; start < t1; start += MF_KCHUNK) {
    const int clen = min(MF_KCHUNK, t1 - start);
    write_staged();
    __syncthreads();
    if (start + MF_KCHUNK < t1)
      issue_loads(start + MF_KCHUNK, t1);

    // S = Q·K^T (wave w: key block w*16..w*16+15); no causal mask in decode
    {
      f32x4_t s_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < MF_D / 32; ++kk) {
        bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
            q_s + a_row * MF_D + ((kk * 32 + a_koff) ^ ((a_row & 15) << 3)));
        bf16x8_t b2 = *reinterpret_cast<const bf16x8_t *>(
            k_s + (wave * 16 + a_row) * MF_D
                + ((kk * 32 + a_koff) ^ ((a_row & 15) << 3)));
        s_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b2, s_acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = c_row0 + r;
        const int col = wave * 16 + c_col;
        s_s[row * SP + col] = (col < clen) ? s_acc[r] * scale : -INFINITY;
      }
    }
    __syncthreads();

    // online softmax (16 threads per row)
    {
      const int row = tid >> 4;
      const int sub = tid & 15;
      float v0 = s_s[row * SP + sub];
      float v1 = s_s[row * SP + sub + 16];
      float v2 = s_s[row * SP + sub + 32];
      float v3 = s_s[row * SP + sub + 48];
      float mymax = fmaxf(fmaxf(v0, v1), fmaxf(v2, v3));
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1)
        mymax = fmaxf(mymax, __shfl_xor(mymax, w, 16));
      const float m_old = m_s[row];
      const float mn = fmaxf(m_old, mymax);
      const float alpha = (m_old == -INFINITY) ? 0.f : __expf(m_old - mn);
      float p0 = (v0 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v0 - mn);
      float p1 = (v1 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v1 - mn);
      float p2 = (v2 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v2 - mn);
      float p3 = (v3 == -INFINITY || mn == -INFINITY) ? 0.f : __expf(v3 - mn);
      float psum = p0 + p1 + p2 + p3;
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1)
        psum += __shfl_xor(psum, w, 16);
      if (sub == 0) {
        l_s[row] = l_s[row] * alpha + psum;
        m_s[row] = mn;
        alpha_s[row] = alpha;
      }
      p_s[row * VP + sub] = f2bf(p0);
      p_s[row * VP + sub + 16] = f2bf(p1);
      p_s[row * VP + sub + 32] = f2bf(p2);
      p_s[row * VP + sub + 48] = f2bf(p3);
    }
    __syncthreads();

    // O rescale + O += P·V (wave w: output cols w*32..w*32+31)
    {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float a = alpha_s[c_row0 + r];
        o_acc0[r] *= a;
        o_acc1[r] *= a;
      }
#pragma unroll
      for (int kk = 0; kk < MF_KCHUNK / 32; ++kk) {
        bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
            p_s + a_row * VP + kk * 32 + a_koff);
        bf16x8_t b0, b1;
#pragma unroll
        for (int t = 0; t < 8; ++t) {
          const bf16 *vrow = v_s + (kk * 32 + a_koff + t) * VR + wave * 32;
          b0[t] = vrow[c_col];
          b1[t] = vrow[16 + c_col];
        }
        o_acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b0, o_acc0, 0, 0, 0);
        o_acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b1, o_acc1, 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // partials out (unnormalized; decode reduce kernel combines splits)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = c_row0 + r;
    if (row >= GQ) continue;
    const int h = hk * GQ + row;
    const long base = ((long)b * Hq + h) * NS + split;
    if (c_col == 0 && wave == 0) {
      part_m[base] = dead ? -INFINITY : m_s[row];
      part_l[base] = dead ? 0.f : l_s[row];
    }
    part_acc[base * MF_D + wave * 32 + c_col] = dead ? 0.f : o_acc0[r];
    part_acc[base * MF_D + wave * 32 + 16 + c_col] = dead ? 0.f : o_acc1[r];
  }
}

// ---------------------------------------------------------------------------
// T12 prefill (the DEFAULT big-prefill kernel, 352 TF measured on the
// bench shape): 128-row Q tiles, 8 waves, each wave OWNS one
// 16-row qblock end to end.  Swapped QK^T (mfma(K, Q)) makes every S row
// lane-local, so the online softmax runs fully in registers (2 shfl_xor
// across the 4 same-qrow lanes) — no s_s/p_s LDS round trip and only TWO
// barriers per chunk (staging).  The P->A-fragment redistribution is the
// guide's T12 permlane dance: pack P to bf16 dwords, then per 32-key
// k-block a pl32swap + pl16swap pair leaves every lane's A-fragment as
// the uniform dword sequence [d0, d1, e0, e1].
//   K: XOR-swizzled [64][128] image (shared staging, as mfma64)
//   V: row-major [64][VR] image; PV B-fragments by conflict-free scalar
//      key-gather (as mfma64)
//   alpha/l cross-lane handoff (softmax lanes hold qrow=lane&15; PV C
//   rows live at (lane>>4)*4+reg) via a tiny per-wave LDS strip.
// Grid: (ntiles128, Hq); block 512.
// ---------------------------------------------------------------------------
#define T12_QT 128

extern "C" __global__ void __launch_bounds__(512, 4)
paged_attn_prefill_t12_kernel(
    bf16 *__restrict__ out, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ tile_q0,
    const int *__restrict__ tile_qn, const int *__restrict__ tile_seq,
    const int *__restrict__ tile_pos0, float scale, int Hq, int Hkv, int BS,
    int MAXB, int GQ) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int hk = h / GQ;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int q0 = tile_q0[tile];
  const int qn = tile_qn[tile];
  const int seq = tile_seq[tile];
  const int pos0 = tile_pos0[tile];
  const int kv_limit = pos0 + qn;

  __shared__ bf16 k_s[MF_KCHUNK * MF_D];     // XOR-swizzled
  __shared__ bf16 v_s[MF_KCHUNK * VR];       // row-major
  __shared__ float alpha_w[8][16];
  __shared__ float l_w[8][16];

  const int qr = lane & 15;                  // this lane's qrow (in qblock)
  const int g = lane >> 4;                   // 16-lane group
  const int row_local = wave * 16 + qr;      // row within the 128-row tile

  // Q B-fragments in registers: lane holds Q[qrow][k = g*8 + t] per kk
  bf16x8_t q_frag[MF_D / 32];
#pragma unroll
  for (int kk = 0; kk < MF_D / 32; ++kk) {
    bf16x8_t v = {};
    if (row_local < qn)
      v = *reinterpret_cast<const bf16x8_t *>(
          q + ((long)(q0 + row_local) * Hq + h) * MF_D + kk * 32 + g * 8);
    q_frag[kk] = v;
  }

  // per-lane online-softmax state for qrow (replicated over the 4
  // same-qrow lanes, deterministically identical)
  float m_run = -INFINITY, l_run = 0.f;
  // O accumulators: 8 d-blocks x f32x4 (C rows = qrows (g*4+r), col = d)
  f32x4_t o_acc[8];
#pragma unroll
  for (int db = 0; db < 8; ++db) o_acc[db] = (f32x4_t){0.f, 0.f, 0.f, 0.f};

  const long panel_stride = (long)Hkv * BS * MF_D;
  const int abs_qrow = min(row_local, qn - 1) + pos0;   // causal bound

  uint4 kreg[2], vreg[2];
  auto issue_loads = [&](int start_, int limit_) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int i = tid + it * 512;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      uint4 kv = make_uint4(0, 0, 0, 0), vv = make_uint4(0, 0, 0, 0);
      const int token = start_ + key;
      if (token < limit_) {
        const long blk = bt[(long)seq * MAXB + token / BS];
        const long off =
            blk * panel_stride + ((long)hk * BS + token % BS) * MF_D + d;
        kv = reinterpret_cast<const uint4 *>(kc + off)[0];
        vv = reinterpret_cast<const uint4 *>(vc + off)[0];
      }
      kreg[it] = kv;
      vreg[it] = vv;
    }
  };
  auto write_staged = [&]() {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int i = tid + it * 512;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      reinterpret_cast<uint4 *>(
          k_s + key * MF_D + (d ^ ((key & 15) << 3)))[0] = kreg[it];
      const uint *vw = reinterpret_cast<const uint *>(&vreg[it]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        reinterpret_cast<uint *>(v_s + key * VR + d)[j] = vw[j];
    }
  };

  issue_loads(0, kv_limit);
  for (int start = 0; start < kv_limit; start += MF_KCHUNK) {
    const int clen = min(MF_KCHUNK, kv_limit - start);
    write_staged();
    __syncthreads();
    if (start + MF_KCHUNK < kv_limit)
      issue_loads(start + MF_KCHUNK, kv_limit);

    // ---- S^T = K·Q^T per 16-key block: C[key][qrow], lane-local rows --
    f32x4_t s_frag[4];
#pragma unroll
    for (int kb = 0; kb < 4; ++kb) {
      f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < MF_D / 32; ++kk) {
        // A = K rows (keys kb*16 + qr), k-slice g*8 (+kk*32), swizzled
        bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
            k_s + (kb * 16 + qr) * MF_D
            + ((kk * 32 + g * 8) ^ ((qr & 15) << 3)));
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, q_frag[kk], acc,
                                                      0, 0, 0);
      }
      s_frag[kb] = acc;
    }

    // ---- in-register online softmax over the 64-key chunk -----------
    // (masked scores, then P, overwrite s_frag in place — the S pipeline
    // must not triple register pressure: vals/pv/s_frag were 48 VGPRs)
    float mymax = -INFINITY;
#pragma unroll
    for (int kb = 0; kb < 4; ++kb) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int key = kb * 16 + g * 4 + r;
        const int token = start + key;
        const bool ok = (token <= abs_qrow) && (key < clen);
        const float sv = ok ? s_frag[kb][r] * scale : -INFINITY;
        s_frag[kb][r] = sv;
        mymax = fmaxf(mymax, sv);
      }
    }
    mymax = fmaxf(mymax, __shfl_xor(mymax, 16));
    mymax = fmaxf(mymax, __shfl_xor(mymax, 32));
    // (T13 defer-max measured perf-neutral here at 3x the rounding error
    // — the rescale pass is only 32 VALU ops — so the exact path stays)
    const float mn = fmaxf(m_run, mymax);
    const float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - mn);
    float psum = 0.f;
#pragma unroll
    for (int kb = 0; kb < 4; ++kb) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = (s_frag[kb][r] == -INFINITY || mn == -INFINITY)
                            ? 0.f : __expf(s_frag[kb][r] - mn);
        psum += p;
        s_frag[kb][r] = p;
      }
    }
    psum += __shfl_xor(psum, 16);
    psum += __shfl_xor(psum, 32);
    l_run = l_run * alpha + psum;
    m_run = mn;
    if (lane < 16) alpha_w[wave][lane] = alpha;   // qrow == lane here

    // ---- O rescale (alpha for PV C rows g*4+r via the wave strip) ----
#pragma unroll
    for (int db = 0; db < 8; ++db) {
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[db][r] *= alpha_w[wave][g * 4 + r];
    }

    // ---- P -> A fragments (permlane dance) + PV ----------------------
    auto packbf = [](float lo, float hi) {
      union { bf16 h; unsigned short u; } a, b;
      a.h = f2bf(lo);
      b.h = f2bf(hi);
      return (uint)a.u | ((uint)b.u << 16);
    };
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      // pack this k-block's two key-blocks (kb0 = 2kk, kb1 = 2kk+1)
      uint d0 = packbf(s_frag[2 * kk][0], s_frag[2 * kk][1]);
      uint d1 = packbf(s_frag[2 * kk][2], s_frag[2 * kk][3]);
      uint e0 = packbf(s_frag[2 * kk + 1][0], s_frag[2 * kk + 1][1]);
      uint e1 = packbf(s_frag[2 * kk + 1][2], s_frag[2 * kk + 1][3]);
      {
        auto r = __builtin_amdgcn_permlane32_swap(d0, e0, false, false);
        d0 = r[0]; e0 = r[1];
      }
      {
        auto r = __builtin_amdgcn_permlane32_swap(d1, e1, false, false);
        d1 = r[0]; e1 = r[1];
      }
      {
        auto r = __builtin_amdgcn_permlane16_swap(d0, e0, false, false);
        d0 = r[0]; e0 = r[1];
      }
      {
        auto r = __builtin_amdgcn_permlane16_swap(d1, e1, false, false);
        d1 = r[0]; e1 = r[1];
      }
      const uint4 av = make_uint4(d0, d1, e0, e1);
      const bf16x8_t a_frag = __builtin_bit_cast(bf16x8_t, av);
#pragma unroll
      for (int db = 0; db < 8; ++db) {
        bf16x8_t b;
#pragma unroll
        for (int t = 0; t < 8; ++t)
          b[t] = v_s[(kk * 32 + g * 8 + t) * VR + db * 16 + qr];
        o_acc[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b,
                                                            o_acc[db], 0, 0, 0);
      }
    }
    __syncthreads();   // next chunk restages k_s / v_s
  }

  if (lane < 16) l_w[wave][lane] = l_run;
  // epilogue: PV C rows = qrows g*4+r of this wave's qblock
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = wave * 16 + g * 4 + r;
    if (row >= qn) continue;
    const float lv = l_w[wave][g * 4 + r];
    const float denom = lv > 0.f ? lv : 1.f;
#pragma unroll
    for (int db = 0; db < 8; ++db)
      out[((long)(q0 + row) * Hq + h) * MF_D + db * 16 + qr] =
          f2bf(o_acc[db][r] / denom);
  }
}

// ---------------------------------------------------------------------------
// EXPERIMENTAL T12-split prefill: small chunks (grammar forced-runs) over
// long cached context.  One 16-row Q tile per workgroup, 8 waves EACH
// walking their own interleaved 16-key chunks with WAVE-LOCAL staging (no
// barriers in the main loop at all) and fully in-register softmax
// (swapped QK^T).  P·V uses the 16-key mfma_f32_16x16x16bf16_1k, whose
// A-fragment k-slice (4 keys per lane) is exactly the lane's own S rows —
// no permlane exchange needed.  Every wave emits its own partial, so the
// host treats waves as 8x more splits: partial layout
//   part_m/part_l: [ntiles, Hq, NS*8, 16];  part_acc: [..., D]
// and the existing paged_attn_prefill_reduce_kernel combines them.
// Grid: (ntiles, Hq, NS); block 512.
// ---------------------------------------------------------------------------
#define TS_CHUNK 16

extern "C" __global__ void __launch_bounds__(512, 4)
paged_attn_prefill_t12_split_kernel(
    float *__restrict__ part_m, float *__restrict__ part_l,
    float *__restrict__ part_acc, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ tile_q0,
    const int *__restrict__ tile_qn, const int *__restrict__ tile_seq,
    const int *__restrict__ tile_pos0, float scale, int Hq, int Hkv, int BS,
    int MAXB, int GQ, int NS) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int split = blockIdx.z;
  const int hk = h / GQ;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int q0 = tile_q0[tile];
  const int qn = tile_qn[tile];
  const int seq = tile_seq[tile];
  const int pos0 = tile_pos0[tile];
  const int kv_limit = pos0 + qn;
  const int chunks = (kv_limit + TS_CHUNK - 1) / TS_CHUNK;
  const int per = (chunks + NS - 1) / NS;
  const int c0 = split * per;                      // this split's chunk range
  const int c1 = min(chunks, (split + 1) * per);

  // wave-local staging buffers: no cross-wave sharing, no loop barriers
  __shared__ bf16 k_w[8][TS_CHUNK * MF_D];         // XOR-swizzled rows
  __shared__ bf16 v_w[8][TS_CHUNK * VR];           // row-major
  __shared__ float alpha_ws[8][16];                // qrow alpha handoff

  const int qr = lane & 15;
  const int g = lane >> 4;

  bf16x8_t q_frag[MF_D / 32];
#pragma unroll
  for (int kk = 0; kk < MF_D / 32; ++kk) {
    bf16x8_t v = {};
    if (qr < qn)
      v = *reinterpret_cast<const bf16x8_t *>(
          q + ((long)(q0 + qr) * Hq + h) * MF_D + kk * 32 + g * 8);
    q_frag[kk] = v;
  }

  float m_run = -INFINITY, l_run = 0.f;
  f32x4_t o_acc[8];
#pragma unroll
  for (int db = 0; db < 8; ++db) o_acc[db] = (f32x4_t){0.f, 0.f, 0.f, 0.f};

  const long panel_stride = (long)Hkv * BS * MF_D;
  const int abs_qrow = min(qr, qn - 1) + pos0;
  bf16 *ks = k_w[wave];
  bf16 *vs = v_w[wave];

  // T14 pipelining, wave-local: chunk c+8's 16 keys ride in registers
  // (4 x uint4 K + V per lane) while chunk c computes
  uint4 kreg[4], vreg[4];
  auto issue_loads = [&](int c) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = lane + it * 64;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      uint4 kv = make_uint4(0, 0, 0, 0), vv = make_uint4(0, 0, 0, 0);
      const int token = c * TS_CHUNK + key;
      if (c < c1 && token < kv_limit) {
        const long blk = bt[(long)seq * MAXB + token / BS];
        const long off =
            blk * panel_stride + ((long)hk * BS + token % BS) * MF_D + d;
        kv = reinterpret_cast<const uint4 *>(kc + off)[0];
        vv = reinterpret_cast<const uint4 *>(vc + off)[0];
      }
      kreg[it] = kv;
      vreg[it] = vv;
    }
  };
  auto write_staged = [&]() {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = lane + it * 64;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      reinterpret_cast<uint4 *>(
          ks + key * MF_D + (d ^ ((key & 15) << 3)))[0] = kreg[it];
      const uint *vw = reinterpret_cast<const uint *>(&vreg[it]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        reinterpret_cast<uint *>(vs + key * VR + d)[j] = vw[j];
    }
  };

  issue_loads(c0 + wave);
  for (int c = c0 + wave; c < c1; c += 8) {
    const int start = c * TS_CHUNK;
    const int clen = min(TS_CHUNK, kv_limit - start);
    write_staged();
    issue_loads(c + 8);
    // wave-local LDS write->read: the compiler's lgkmcnt wait suffices

    // S^T = K·Q^T (one 16-key block)
    f32x4_t s_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < MF_D / 32; ++kk) {
      bf16x8_t a = *reinterpret_cast<const bf16x8_t *>(
          ks + qr * MF_D + ((kk * 32 + g * 8) ^ ((qr & 15) << 3)));
      s_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, q_frag[kk], s_acc,
                                                      0, 0, 0);
    }

    // in-register online softmax over the 16-key chunk
    float mymax = -INFINITY;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = g * 4 + r;
      const int token = start + key;
      const bool ok = (token <= abs_qrow) && (key < clen);
      const float sv = ok ? s_acc[r] * scale : -INFINITY;
      s_acc[r] = sv;
      mymax = fmaxf(mymax, sv);
    }
    mymax = fmaxf(mymax, __shfl_xor(mymax, 16));
    mymax = fmaxf(mymax, __shfl_xor(mymax, 32));
    const float mn = fmaxf(m_run, mymax);
    const float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - mn);
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float p = (s_acc[r] == -INFINITY || mn == -INFINITY)
                          ? 0.f : __expf(s_acc[r] - mn);
      psum += p;
      s_acc[r] = p;
    }
    psum += __shfl_xor(psum, 16);
    psum += __shfl_xor(psum, 32);
    l_run = l_run * alpha + psum;
    m_run = mn;
    // softmax state lives at qrow = lane&15; the PV C rows are qrows
    // g*4+r — hand the alphas across lanes through the wave strip
    if (lane < 16) alpha_ws[wave][lane] = alpha;

    // A-fragment for the 16-key PV MFMA: exactly this lane's 4 P values
    bf16x4_t a16;
#pragma unroll
    for (int r = 0; r < 4; ++r) a16[r] = f2bf(s_acc[r]);

#pragma unroll
    for (int db = 0; db < 8; ++db) {
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[db][r] *= alpha_ws[wave][g * 4 + r];
      bf16x4_t b;
#pragma unroll
      for (int t = 0; t < 4; ++t)
        b[t] = vs[(g * 4 + t) * VR + db * 16 + qr];
      o_acc[db] = __builtin_amdgcn_mfma_f32_16x16x16bf16_1k(a16, b,
                                                            o_acc[db], 0, 0, 0);
    }
  }

  // per-wave partials at split index (split*8 + wave)
  const bool dead = (c0 + wave >= c1);
  const long base = (((long)tile * Hq + h) * (NS * 8) + split * 8 + wave)
                    * QT;
  // m/l per qrow live at lanes 0-15 (their own qrow), duplicated on the
  // other lane groups — lane<16 writes them
  if (lane < 16) {
    part_m[base + lane] = dead ? -INFINITY : m_run;
    part_l[base + lane] = dead ? 0.f : l_run;
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = g * 4 + r;             // PV C row = qrow
#pragma unroll
    for (int db = 0; db < 8; ++db)
      part_acc[(base + row) * MF_D + db * 16 + qr] =
          dead ? 0.f : o_acc[db][r];
  }
}

// ---------------------------------------------------------------------------
// EXPERIMENTAL T12W prefill: the T12 structure on the 32x32x16 MFMA
// (double the flops per instruction: guide µbench 2382 vs 2075 TF).
// 4 waves per WG, wave owns a 32-row qblock (same 128-row tiles as T12).
// Swapped QK^T gives lane-local S rows (qrow = lane&31, keys by the
// 32x32 C map (reg&3)+8*(reg>>2)+4*(lane>>5)); softmax fully in
// registers (one shfl_xor(32)); the PV A-fragments come from FOUR
// permlane32_swaps per 32-key block:
//   swap(j0,j2) -> [ks=0 D0, ks=0 D2]; swap(j1,j3) -> [D1, D3]
//   swap(j4,j6) -> [ks=1 D0, D2];      swap(j5,j7) -> [D1, D3]
// (dword j = cvt_pk of P at regs 2j, 2j+1 — adjacent keys by the C map).
// Q per wave lives in q_s (XOR-swizzled [128][128]); K/V staged as in
// the 4-wave kernels; 2 barriers per 64-key chunk.
// Grid: (ntiles128, Hq); block 256.
// ---------------------------------------------------------------------------
typedef float f32x16_t __attribute__((ext_vector_type(16)));

extern "C" __global__ void __launch_bounds__(256)
paged_attn_prefill_t12w_kernel(
    bf16 *__restrict__ out, const bf16 *__restrict__ q,
    const bf16 *__restrict__ kc, const bf16 *__restrict__ vc,
    const int *__restrict__ bt, const int *__restrict__ tile_q0,
    const int *__restrict__ tile_qn, const int *__restrict__ tile_seq,
    const int *__restrict__ tile_pos0, float scale, int Hq, int Hkv, int BS,
    int MAXB, int GQ) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int hk = h / GQ;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int q0 = tile_q0[tile];
  const int qn = tile_qn[tile];
  const int seq = tile_seq[tile];
  const int pos0 = tile_pos0[tile];
  const int kv_limit = pos0 + qn;

  __shared__ bf16 q_s[T12_QT * MF_D];        // XOR-swizzled 128-row Q
  __shared__ bf16 k_s[MF_KCHUNK * MF_D];     // XOR-swizzled
  __shared__ bf16 v_s[MF_KCHUNK * VR];       // row-major
  __shared__ float alpha_w[4][32];
  __shared__ float l_w[4][32];

  const int qr = lane & 31;                  // lane's qrow within qblock
  const int hi = lane >> 5;
  const int row_local = wave * 32 + qr;      // row within the 128-row tile

  // stage the Q tile (swizzled, all 4 waves cooperate)
  for (int i = tid; i < T12_QT * MF_D / 8; i += 256) {
    const int r = (i * 8) / MF_D, c = (i * 8) % MF_D;
    uint4 val = make_uint4(0, 0, 0, 0);
    if (r < qn)
      val = reinterpret_cast<const uint4 *>(
          q + ((long)(q0 + r) * Hq + h) * MF_D + c)[0];
    reinterpret_cast<uint4 *>(
        q_s + r * MF_D + (c ^ ((r & 15) << 3)))[0] = val;
  }

  float m_run = -INFINITY, l_run = 0.f;
  f32x16_t o_acc[4];                         // 4 d-tiles of 32 cols
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) o_acc[dt] = (f32x16_t){};

  const long panel_stride = (long)Hkv * BS * MF_D;
  const int abs_qrow = min(row_local, qn - 1) + pos0;

  uint4 kreg[4], vreg[4];
  auto issue_loads = [&](int start_, int limit_) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = tid + it * 256;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      uint4 kv = make_uint4(0, 0, 0, 0), vv = make_uint4(0, 0, 0, 0);
      const int token = start_ + key;
      if (token < limit_) {
        const long blk = bt[(long)seq * MAXB + token / BS];
        const long off =
            blk * panel_stride + ((long)hk * BS + token % BS) * MF_D + d;
        kv = reinterpret_cast<const uint4 *>(kc + off)[0];
        vv = reinterpret_cast<const uint4 *>(vc + off)[0];
      }
      kreg[it] = kv;
      vreg[it] = vv;
    }
  };
  auto write_staged = [&]() {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int i = tid + it * 256;
      const int key = (i * 8) / MF_D, d = (i * 8) % MF_D;
      reinterpret_cast<uint4 *>(
          k_s + key * MF_D + (d ^ ((key & 15) << 3)))[0] = kreg[it];
      const uint *vw = reinterpret_cast<const uint *>(&vreg[it]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        reinterpret_cast<uint *>(v_s + key * VR + d)[j] = vw[j];
    }
  };

  auto packbf = [](float lo, float hif) {
    union { bf16 hh; unsigned short u; } a, b;
    a.hh = f2bf(lo);
    b.hh = f2bf(hif);
    return (uint)a.u | ((uint)b.u << 16);
  };

  issue_loads(0, kv_limit);
  for (int start = 0; start < kv_limit; start += MF_KCHUNK) {
    const int clen = min(MF_KCHUNK, kv_limit - start);
    write_staged();
    __syncthreads();
    if (start + MF_KCHUNK < kv_limit)
      issue_loads(start + MF_KCHUNK, kv_limit);

    // ---- S^T = K·Q^T: 2 key-blocks of 32, k-loop over D in 16s -------
    f32x16_t s_acc[2];
    s_acc[0] = (f32x16_t){};
    s_acc[1] = (f32x16_t){};
#pragma unroll
    for (int kk = 0; kk < MF_D / 16; ++kk) {
      const int doff = (kk * 16 + hi * 8);
      bf16x8_t qf = *reinterpret_cast<const bf16x8_t *>(
          q_s + (wave * 32 + qr) * MF_D
          + (doff ^ (((wave * 32 + qr) & 15) << 3)));
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
        bf16x8_t kf = *reinterpret_cast<const bf16x8_t *>(
            k_s + (kb * 32 + qr) * MF_D + (doff ^ ((qr & 15) << 3)));
        s_acc[kb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf,
                                                            s_acc[kb], 0, 0, 0);
      }
    }

    // ---- in-register online softmax over the 64-key chunk ------------
    float mymax = -INFINITY;
#pragma unroll
    for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int key = kb * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const int token = start + key;
        const bool ok = (token <= abs_qrow) && (key < clen);
        const float sv = ok ? s_acc[kb][r] * scale : -INFINITY;
        s_acc[kb][r] = sv;
        mymax = fmaxf(mymax, sv);
      }
    }
    mymax = fmaxf(mymax, __shfl_xor(mymax, 32));
    const float mn = fmaxf(m_run, mymax);
    const float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - mn);
    float psum = 0.f;
#pragma unroll
    for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const float p = (s_acc[kb][r] == -INFINITY || mn == -INFINITY)
                            ? 0.f : __expf(s_acc[kb][r] - mn);
        psum += p;
        s_acc[kb][r] = p;
      }
    }
    psum += __shfl_xor(psum, 32);
    l_run = l_run * alpha + psum;
    m_run = mn;
    if (lane < 32) alpha_w[wave][lane] = alpha;   // qrow == lane here

    // ---- O rescale (alphas for the C row map via the wave strip) -----
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
      for (int r = 0; r < 16; ++r)
        o_acc[dt][r] *= alpha_w[wave][(r & 3) + 8 * (r >> 2) + 4 * hi];
    }

    // ---- P -> A fragments (4 swaps per 32-key block) + PV ------------
#pragma unroll
    for (int kb = 0; kb < 2; ++kb) {
      uint j[8];
#pragma unroll
      for (int m = 0; m < 8; ++m)
        j[m] = packbf(s_acc[kb][2 * m], s_acc[kb][2 * m + 1]);
      // swap(j0,j2): ks=0 D0/D2; swap(j1,j3): ks=0 D1/D3;
      // swap(j4,j6): ks=1 D0/D2; swap(j5,j7): ks=1 D1/D3
      uint a0d0, a0d1, a0d2, a0d3, a1d0, a1d1, a1d2, a1d3;
      {
        auto r = __builtin_amdgcn_permlane32_swap(j[0], j[2], false, false);
        a0d0 = r[0]; a0d2 = r[1];
      }
      {
        auto r = __builtin_amdgcn_permlane32_swap(j[1], j[3], false, false);
        a0d1 = r[0]; a0d3 = r[1];
      }
      {
        auto r = __builtin_amdgcn_permlane32_swap(j[4], j[6], false, false);
        a1d0 = r[0]; a1d2 = r[1];
      }
      {
        auto r = __builtin_amdgcn_permlane32_swap(j[5], j[7], false, false);
        a1d1 = r[0]; a1d3 = r[1];
      }
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const uint4 av = half
            ? make_uint4(a1d0, a1d1, a1d2, a1d3)
            : make_uint4(a0d0, a0d1, a0d2, a0d3);
        const bf16x8_t a_frag = __builtin_bit_cast(bf16x8_t, av);
        const int ksbase = kb * 32 + half * 16;      // key-slice origin
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          bf16x8_t b;
#pragma unroll
          for (int t = 0; t < 8; ++t)
            b[t] = v_s[(ksbase + hi * 8 + t) * VR + dt * 32 + qr];
          o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_frag, b, o_acc[dt], 0, 0, 0);
        }
      }
    }
    __syncthreads();   // next chunk restages k_s / v_s
  }

  if (lane < 32) l_w[wave][lane] = l_run;
  // epilogue: C rows = (r&3)+8*(r>>2)+4*hi of this wave's qblock
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int rowq = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const int row = wave * 32 + rowq;
    if (row >= qn) continue;
    const float lv = l_w[wave][rowq];
    const float denom = lv > 0.f ? lv : 1.f;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      out[((long)(q0 + row) * Hq + h) * MF_D + dt * 32 + qr] =
          f2bf(o_acc[dt][r] / denom);
  }
}
