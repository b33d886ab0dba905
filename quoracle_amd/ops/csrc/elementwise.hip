// Fused elementwise / normalization kernels for the MI355X inference path.
//
// These ops are HBM-bandwidth-bound (MI355X: ~8 TB/s peak, ~6.3 achievable);
// the design rule (cdna_hip_programming.md §2, common-mistake #2) is bf16x8
// vector I/O via uint4 and fusion of the residual add into the norm so each
// tensor crosses HBM exactly once.

#include "common.h"

// ---------------------------------------------------------------------------
// rmsnorm_fused: y = rmsnorm(x [+ residual]) * w ; residual updated in place.
//   x:        [rows, n] bf16
//   residual: [rows, n] bf16 or nullptr; on exit holds (x + residual)
//   w:        [n] bf16
//   y:        [rows, n] bf16
// One block per row; n must be a multiple of 8.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
rmsnorm_fused_kernel(bf16 *__restrict__ y, const bf16 *__restrict__ x,
                     bf16 *__restrict__ residual, const bf16 *__restrict__ w,
                     int n, float eps) {
  __shared__ float scratch[8];
  __shared__ float s_inv;
  const long row = blockIdx.x;
  const bf16 *xr = x + row * (long)n;
  bf16 *rr = residual ? residual + row * (long)n : nullptr;
  bf16 *yr = y + row * (long)n;

  const int vecs = n / 8;  // uint4 = 8 bf16
  float ssq = 0.f;
  // First pass: (optional residual add, written back) + sum of squares.
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    uint4 xv = reinterpret_cast<const uint4 *>(xr)[i];
    float vals[8];
    unpack_bf16x2(xv.x, vals[0], vals[1]);
    unpack_bf16x2(xv.y, vals[2], vals[3]);
    unpack_bf16x2(xv.z, vals[4], vals[5]);
    unpack_bf16x2(xv.w, vals[6], vals[7]);
    if (rr) {
      uint4 rv = reinterpret_cast<const uint4 *>(rr)[i];
      float rvals[8];
      unpack_bf16x2(rv.x, rvals[0], rvals[1]);
      unpack_bf16x2(rv.y, rvals[2], rvals[3]);
      unpack_bf16x2(rv.z, rvals[4], rvals[5]);
      unpack_bf16x2(rv.w, rvals[6], rvals[7]);
#pragma unroll
      for (int k = 0; k < 8; ++k) vals[k] += rvals[k];
      uint4 out;
      out.x = pack_bf16x2(vals[0], vals[1]);
      out.y = pack_bf16x2(vals[2], vals[3]);
      out.z = pack_bf16x2(vals[4], vals[5]);
      out.w = pack_bf16x2(vals[6], vals[7]);
      reinterpret_cast<uint4 *>(rr)[i] = out;
    }
#pragma unroll
    for (int k = 0; k < 8; ++k) ssq += vals[k] * vals[k];
  }
  float total = block_sum(ssq, scratch);
  if (threadIdx.x == 0) s_inv = rsqrtf(total / (float)n + eps);
  __syncthreads();
  const float inv = s_inv;

  // Second pass: normalize * weight.  Re-read the (residual-updated) row —
  // for n=4096 the row is L1/L2-hot, so this costs no HBM traffic.
  const bf16 *src = rr ? (const bf16 *)rr : xr;
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    uint4 xv = reinterpret_cast<const uint4 *>(src)[i];
    uint4 wv = reinterpret_cast<const uint4 *>(w)[i];
    float vals[8], ws[8];
    unpack_bf16x2(xv.x, vals[0], vals[1]);
    unpack_bf16x2(xv.y, vals[2], vals[3]);
    unpack_bf16x2(xv.z, vals[4], vals[5]);
    unpack_bf16x2(xv.w, vals[6], vals[7]);
    unpack_bf16x2(wv.x, ws[0], ws[1]);
    unpack_bf16x2(wv.y, ws[2], ws[3]);
    unpack_bf16x2(wv.z, ws[4], ws[5]);
    unpack_bf16x2(wv.w, ws[6], ws[7]);
    uint4 out;
    out.x = pack_bf16x2(vals[0] * inv * ws[0], vals[1] * inv * ws[1]);
    out.y = pack_bf16x2(vals[2] * inv * ws[2], vals[3] * inv * ws[3]);
    out.z = pack_bf16x2(vals[4] * inv * ws[4], vals[5] * inv * ws[5]);
    out.w = pack_bf16x2(vals[6] * inv * ws[6], vals[7] * inv * ws[7]);
    reinterpret_cast<uint4 *>(yr)[i] = out;
  }
}

// ---------------------------------------------------------------------------
// swiglu: out[t, j] = silu(gu[t, j]) * gu[t, inter + j]
//   gu:  [rows, 2*inter] bf16 (gate | up, the fused gate_up projection)
//   out: [rows, inter] bf16
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
swiglu_kernel(bf16 *__restrict__ out, const bf16 *__restrict__ gu,
              int inter) {
  const long row = blockIdx.x;
  const bf16 *g = gu + row * (long)(2 * inter);
  const bf16 *u = g + inter;
  bf16 *o = out + row * (long)inter;
  const int vecs = inter / 8;
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    uint4 gv = reinterpret_cast<const uint4 *>(g)[i];
    uint4 uv = reinterpret_cast<const uint4 *>(u)[i];
    float gs[8], us[8];
    unpack_bf16x2(gv.x, gs[0], gs[1]);
    unpack_bf16x2(gv.y, gs[2], gs[3]);
    unpack_bf16x2(gv.z, gs[4], gs[5]);
    unpack_bf16x2(gv.w, gs[6], gs[7]);
    unpack_bf16x2(uv.x, us[0], us[1]);
    unpack_bf16x2(uv.y, us[2], us[3]);
    unpack_bf16x2(uv.z, us[4], us[5]);
    unpack_bf16x2(uv.w, us[6], us[7]);
    float r[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float x = gs[k];
      float silu = x / (1.f + __expf(-x));
      r[k] = silu * us[k];
    }
    uint4 ov;
    ov.x = pack_bf16x2(r[0], r[1]);
    ov.y = pack_bf16x2(r[2], r[3]);
    ov.z = pack_bf16x2(r[4], r[5]);
    ov.w = pack_bf16x2(r[6], r[7]);
    reinterpret_cast<uint4 *>(o)[i] = ov;
  }
}

// ---------------------------------------------------------------------------
// rope_inplace: rotate-half RoPE applied to q and k in place.
//   q: [T, Hq, D] bf16;  k: [T, Hk, D] bf16;  pos: [T] int32
// Llama convention: pair (d, d + D/2); freq = theta^(-2d/D).
// Grid: (T, Hq + Hk); block: D/2 threads (D <= 256).
// ---------------------------------------------------------------------------
extern "C" __global__ void rope_inplace_kernel(
    bf16 *__restrict__ q, bf16 *__restrict__ k,
    const int *__restrict__ pos, int Hq, int Hk, int D, float theta) {
  const long t = blockIdx.x;
  const int h = blockIdx.y;
  const int d = threadIdx.x;         // 0 .. D/2-1
  const int half = D / 2;
  if (d >= half) return;
  bf16 *base = (h < Hq) ? q + (t * Hq + h) * (long)D
                        : k + (t * Hk + (h - Hq)) * (long)D;
  const float p = (float)pos[t];
  const float freq = __powf(theta, -2.f * (float)d / (float)D);
  float c, s;
  __sincosf(p * freq, &s, &c);
  float x0 = bf2f(base[d]);
  float x1 = bf2f(base[d + half]);
  base[d] = f2bf(x0 * c - x1 * s);
  base[d + half] = f2bf(x0 * s + x1 * c);
}

// ---------------------------------------------------------------------------
// kv_append: scatter new K/V rows into the paged cache.
//   kcache/vcache: [num_blocks, Hk, block_size, D] bf16
//   k/v:           [T, Hk, D] bf16
//   slots:         [T] int32 — global slot = block * block_size + offset
// Grid: (T, Hk); block: D threads.
// ---------------------------------------------------------------------------
extern "C" __global__ void kv_append_kernel(
    bf16 *__restrict__ kcache, bf16 *__restrict__ vcache,
    const bf16 *__restrict__ k, const bf16 *__restrict__ v,
    const int *__restrict__ slots, int Hk, int block_size, int D) {
  const long t = blockIdx.x;
  const int h = blockIdx.y;
  const int d = threadIdx.x;
  const int slot = slots[t];
  if (slot < 0 || d >= D) return;
  const long blk = slot / block_size;
  const int off = slot % block_size;
  const long dst = ((blk * gridDim.y + h) * (long)block_size + off) * D + d;
  kcache[dst] = k[(t * Hk + h) * (long)D + d];
  vcache[dst] = v[(t * Hk + h) * (long)D + d];
}

// ---------------------------------------------------------------------------
// cosine_sim_matrix: out[i][j] = cos(x[i], x[j]) for the consensus vote.
//   x: [N, D] f32 ;  out: [N, N] f32
// One block per (i, j) pair with j >= i (symmetric); zero-norm rows -> 0.
// N is tiny (pool size); this replaces the reference's serial CPU loop
// (reference: aggregator.ex:335-351) with one fused kernel.
// ---------------------------------------------------------------------------
extern "C" __global__ void cosine_sim_kernel(
    float *__restrict__ out, const float *__restrict__ x, int N, int D) {
  __shared__ float scratch[8];
  const int i = blockIdx.x;
  const int j = blockIdx.y;
  if (j < i) return;
  float dot = 0.f, mi = 0.f, mj = 0.f;
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
    float a = x[i * (long)D + d];
    float b = x[j * (long)D + d];
    dot += a * b;
    mi += a * a;
    mj += b * b;
  }
  // three block reductions share the scratch sequentially
  dot = block_sum(dot, scratch);
  mi = block_sum(mi, scratch);
  mj = block_sum(mj, scratch);
  if (threadIdx.x == 0) {
    float denom = sqrtf(mi) * sqrtf(mj);
    float sim = denom > 0.f ? dot / denom : 0.f;
    out[i * (long)N + j] = sim;
    out[j * (long)N + i] = sim;
  }
}

// ---------------------------------------------------------------------------
// gather_rows_bf16: out[i] = src[rows[i]] — used for the embedding lookup.
//   src: [V, n] bf16; rows: [T] int32; out: [T, n] bf16 (n % 8 == 0)
// ---------------------------------------------------------------------------
extern "C" __global__ void gather_rows_kernel(
    bf16 *__restrict__ out, const bf16 *__restrict__ src,
    const int *__restrict__ rows, int n) {
  const long t = blockIdx.x;
  const long r = rows[t];
  const uint4 *s = reinterpret_cast<const uint4 *>(src + r * (long)n);
  uint4 *o = reinterpret_cast<uint4 *>(out + t * (long)n);
  for (int i = threadIdx.x; i < n / 8; i += blockDim.x) o[i] = s[i];
}

// ---------------------------------------------------------------------------
// split_qkv: one pass splitting the fused QKV GEMM output into contiguous
// q/k/v tensors (replaces three narrow().contiguous() dispatch+copy chains
// per layer in the forward driver).  Row layout: [q_dim | kv_dim | kv_dim].
//   qkv: [T, q_dim + 2*kv_dim] bf16;  q: [T, q_dim];  k/v: [T, kv_dim]
// Grid: (T); block 256.  Dims % 8 == 0.
// ---------------------------------------------------------------------------
extern "C" __global__ void split_qkv_kernel(
    bf16 *__restrict__ q, bf16 *__restrict__ k, bf16 *__restrict__ v,
    const bf16 *__restrict__ qkv, int q_dim, int kv_dim) {
  const long t = blockIdx.x;
  const long row = t * (long)(q_dim + 2 * kv_dim);
  const uint4 *src = reinterpret_cast<const uint4 *>(qkv + row);
  uint4 *qo = reinterpret_cast<uint4 *>(q + t * (long)q_dim);
  uint4 *ko = reinterpret_cast<uint4 *>(k + t * (long)kv_dim);
  uint4 *vo = reinterpret_cast<uint4 *>(v + t * (long)kv_dim);
  const int qv = q_dim / 8, kv8 = kv_dim / 8;
  for (int i = threadIdx.x; i < qv + 2 * kv8; i += blockDim.x) {
    const uint4 val = src[i];
    if (i < qv) qo[i] = val;
    else if (i < qv + kv8) ko[i - qv] = val;
    else vo[i - qv - kv8] = val;
  }
}
