// Torch bindings for the quoracle_amd HIP kernels (gfx950).
//
// Single translation unit: the kernel .hip files are included directly so no
// relocatable device code is needed.  Built in-tree by quoracle_amd/ops/build.py
// with `hipcc -x hip --offload-arch=gfx950` (see __graft_entry__.build()).

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

#include <stdexcept>
#include <string>

#include "attention.hip"
#include "attention_mfma.hip"
#include "elementwise.hip"

namespace {

#define CHECK_GPU(t)                                                     \
  TORCH_CHECK((t).is_cuda(), #t " must be on the GPU");                  \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

inline bf16 *bf16_ptr(torch::Tensor &t) {
  return reinterpret_cast<bf16 *>(t.data_ptr());
}
inline const bf16 *bf16_cptr(const torch::Tensor &t) {
  return reinterpret_cast<const bf16 *>(t.data_ptr());
}

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void rmsnorm_fused(torch::Tensor y, torch::Tensor x,
                   c10::optional<torch::Tensor> residual, torch::Tensor w,
                   double eps) {
  CHECK_GPU(y);
  CHECK_GPU(x);
  CHECK_GPU(w);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  const int n = x.size(-1);
  TORCH_CHECK(n % 8 == 0, "hidden size must be a multiple of 8");
  const long rows = x.numel() / n;
  bf16 *res_ptr = nullptr;
  if (residual.has_value()) {
    CHECK_GPU(residual.value());
    res_ptr = bf16_ptr(residual.value());
  }
  hipLaunchKernelGGL(rmsnorm_fused_kernel, dim3((unsigned)rows), dim3(256), 0,
                     current_stream(), bf16_ptr(y), bf16_cptr(x), res_ptr,
                     bf16_cptr(w), n, (float)eps);
  HIP_CHECK_KERNEL();
}

void swiglu(torch::Tensor out, torch::Tensor gate_up) {
  CHECK_GPU(out);
  CHECK_GPU(gate_up);
  const int inter = out.size(-1);
  TORCH_CHECK(gate_up.size(-1) == 2 * inter, "gate_up must be [rows, 2*inter]");
  TORCH_CHECK(inter % 8 == 0, "inter must be a multiple of 8");
  const long rows = out.numel() / inter;
  hipLaunchKernelGGL(swiglu_kernel, dim3((unsigned)rows), dim3(256), 0,
                     current_stream(), bf16_ptr(out), bf16_cptr(gate_up),
                     inter);
  HIP_CHECK_KERNEL();
}

void split_qkv(torch::Tensor q, torch::Tensor k, torch::Tensor v,
               torch::Tensor qkv) {
  CHECK_GPU(q);
  CHECK_GPU(k);
  CHECK_GPU(v);
  CHECK_GPU(qkv);
  const long T = qkv.size(0);
  const int q_dim = q.numel() / T;
  const int kv_dim = k.numel() / T;
  TORCH_CHECK(qkv.size(1) == q_dim + 2 * kv_dim, "qkv row layout");
  TORCH_CHECK(q_dim % 8 == 0 && kv_dim % 8 == 0, "dims % 8");
  if (T == 0) return;
  hipLaunchKernelGGL(split_qkv_kernel, dim3((unsigned)T), dim3(256), 0,
                     current_stream(), bf16_ptr(q), bf16_ptr(k), bf16_ptr(v),
                     bf16_cptr(qkv), q_dim, kv_dim);
  HIP_CHECK_KERNEL();
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor pos,
                  double theta) {
  CHECK_GPU(q);
  CHECK_GPU(k);
  CHECK_GPU(pos);
  TORCH_CHECK(pos.scalar_type() == torch::kInt32, "pos must be int32");
  const int T = q.size(0);
  const int Hq = q.size(1);
  const int Hk = k.size(1);
  const int D = q.size(2);
  TORCH_CHECK(D % 2 == 0 && D <= 512, "bad head dim");
  if (T == 0) return;
  hipLaunchKernelGGL(rope_inplace_kernel, dim3(T, Hq + Hk), dim3(D / 2), 0,
                     current_stream(), bf16_ptr(q), bf16_ptr(k),
                     pos.data_ptr<int>(), Hq, Hk, D, (float)theta);
  HIP_CHECK_KERNEL();
}

void kv_append(torch::Tensor kcache, torch::Tensor vcache, torch::Tensor k,
               torch::Tensor v, torch::Tensor slots) {
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(k);
  CHECK_GPU(v);
  CHECK_GPU(slots);
  TORCH_CHECK(slots.scalar_type() == torch::kInt32, "slots must be int32");
  const int T = k.size(0);
  const int Hk = k.size(1);
  const int D = k.size(2);
  const int BS = kcache.size(2);
  if (T == 0) return;
  hipLaunchKernelGGL(kv_append_kernel, dim3(T, Hk), dim3(D), 0,
                     current_stream(), bf16_ptr(kcache), bf16_ptr(vcache),
                     bf16_cptr(k), bf16_cptr(v), slots.data_ptr<int>(), Hk, BS,
                     D);
  HIP_CHECK_KERNEL();
}

void paged_attn_decode(torch::Tensor out, torch::Tensor q,
                       torch::Tensor kcache, torch::Tensor vcache,
                       torch::Tensor block_tables, torch::Tensor ctx_lens,
                       double scale) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  CHECK_GPU(ctx_lens);
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  TORCH_CHECK(Hq % Hkv == 0 && GQ <= MAX_GQ, "unsupported GQA ratio");
  TORCH_CHECK(D <= DECODE_BLOCK, "head dim too large");
  if (B == 0) return;
  hipLaunchKernelGGL(paged_attn_decode_kernel, dim3(B, Hkv),
                     dim3(DECODE_BLOCK), 0, current_stream(), bf16_ptr(out),
                     bf16_cptr(q), bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), ctx_lens.data_ptr<int>(),
                     (float)scale, Hq, Hkv, D, BS, MAXB, GQ);
  HIP_CHECK_KERNEL();
}

void paged_attn_decode_split(torch::Tensor out, torch::Tensor q,
                             torch::Tensor kcache, torch::Tensor vcache,
                             torch::Tensor block_tables,
                             torch::Tensor ctx_lens, double scale,
                             torch::Tensor part_m, torch::Tensor part_l,
                             torch::Tensor part_acc) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  CHECK_GPU(ctx_lens);
  CHECK_GPU(part_m);
  CHECK_GPU(part_l);
  CHECK_GPU(part_acc);
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  const int NS = part_m.size(2);
  TORCH_CHECK(Hq % Hkv == 0 && GQ <= MAX_GQ, "unsupported GQA ratio");
  TORCH_CHECK(D <= DECODE_BLOCK, "head dim too large");
  TORCH_CHECK(part_m.size(0) >= B && part_m.size(1) == Hq, "workspace shape");
  TORCH_CHECK(part_acc.size(3) == D, "workspace D");
  if (B == 0) return;
  hipLaunchKernelGGL(paged_attn_decode_split_kernel, dim3(B, Hkv, NS),
                     dim3(DECODE_BLOCK), 0, current_stream(),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), bf16_cptr(q),
                     bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), ctx_lens.data_ptr<int>(),
                     (float)scale, Hq, Hkv, D, BS, MAXB, GQ, NS);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(paged_attn_decode_reduce_kernel, dim3(B, Hq),
                     dim3(DECODE_BLOCK), 0, current_stream(), bf16_ptr(out),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), Hq, D, NS);
  HIP_CHECK_KERNEL();
}

void paged_attn_decode_split2(torch::Tensor out, torch::Tensor q,
                              torch::Tensor kcache, torch::Tensor vcache,
                              torch::Tensor block_tables,
                              torch::Tensor ctx_lens, double scale,
                              torch::Tensor part_m, torch::Tensor part_l,
                              torch::Tensor part_acc) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  CHECK_GPU(ctx_lens);
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  const int NS = part_m.size(2);
  TORCH_CHECK(Hq % Hkv == 0 && GQ <= MAX_GQ, "unsupported GQA ratio");
  TORCH_CHECK(D <= 128, "head dim too large");
  TORCH_CHECK(256 % BS == 0 && BS <= 256, "block size must divide 256");
  if (B == 0) return;
  hipLaunchKernelGGL(paged_attn_decode_split2_kernel, dim3(B, Hkv, NS),
                     dim3(256), 0, current_stream(),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), bf16_cptr(q),
                     bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), ctx_lens.data_ptr<int>(),
                     (float)scale, Hq, Hkv, D, BS, MAXB, GQ, NS);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(paged_attn_decode_reduce_kernel, dim3(B, Hq),
                     dim3(DECODE_BLOCK), 0, current_stream(), bf16_ptr(out),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), Hq, D, NS);
  HIP_CHECK_KERNEL();
}

void paged_attn_decode_mfma(torch::Tensor out, torch::Tensor q,
                            torch::Tensor kcache, torch::Tensor vcache,
                            torch::Tensor block_tables,
                            torch::Tensor ctx_lens, double scale,
                            torch::Tensor part_m, torch::Tensor part_l,
                            torch::Tensor part_acc) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  CHECK_GPU(ctx_lens);
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  const int NS = part_m.size(2);
  TORCH_CHECK(Hq % Hkv == 0 && GQ <= MF_QT, "unsupported GQA ratio");
  TORCH_CHECK(D == MF_D, "MFMA decode requires head dim 128");
  TORCH_CHECK(part_m.size(0) >= B && part_m.size(1) == Hq, "workspace shape");
  TORCH_CHECK(part_acc.size(3) == D, "workspace D");
  if (B == 0) return;
  hipLaunchKernelGGL(paged_attn_decode_mfma_kernel, dim3(B, Hkv, NS),
                     dim3(256), 0, current_stream(),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), bf16_cptr(q),
                     bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), ctx_lens.data_ptr<int>(),
                     (float)scale, Hq, Hkv, BS, MAXB, GQ, NS);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(paged_attn_decode_reduce_kernel, dim3(B, Hq),
                     dim3(DECODE_BLOCK), 0, current_stream(), bf16_ptr(out),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), Hq, D, NS);
  HIP_CHECK_KERNEL();
}

void paged_attn_prefill(torch::Tensor out, torch::Tensor q,
                        torch::Tensor kcache, torch::Tensor vcache,
                        torch::Tensor block_tables, torch::Tensor tile_q0,
                        torch::Tensor tile_qn, torch::Tensor tile_seq,
                        torch::Tensor tile_pos0, double scale) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  const int ntiles = tile_q0.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  if (ntiles == 0) return;
  hipLaunchKernelGGL(paged_attn_prefill_kernel, dim3(ntiles, Hq),
                     dim3(DECODE_BLOCK), 0, current_stream(), bf16_ptr(out),
                     bf16_cptr(q), bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), tile_seq.data_ptr<int>(),
                     tile_pos0.data_ptr<int>(), (float)scale, Hq, Hkv, D, BS,
                     MAXB, GQ);
  HIP_CHECK_KERNEL();
}

void paged_attn_prefill_split(torch::Tensor out, torch::Tensor q,
                              torch::Tensor kcache, torch::Tensor vcache,
                              torch::Tensor block_tables,
                              torch::Tensor tile_q0, torch::Tensor tile_qn,
                              torch::Tensor tile_seq, torch::Tensor tile_pos0,
                              double scale, torch::Tensor part_m,
                              torch::Tensor part_l, torch::Tensor part_acc) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  CHECK_GPU(part_m);
  CHECK_GPU(part_l);
  CHECK_GPU(part_acc);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  const int ntiles = tile_q0.size(0);
  const int NS = part_m.size(2);
  TORCH_CHECK(Hq % Hkv == 0, "bad GQA ratio");
  TORCH_CHECK(D <= DECODE_BLOCK, "head dim too large");
  TORCH_CHECK(part_m.size(0) >= ntiles && part_m.size(1) == Hq &&
              part_m.size(3) == QT, "workspace shape");
  TORCH_CHECK(part_acc.size(4) == D, "workspace D");
  if (ntiles == 0) return;
  hipLaunchKernelGGL(paged_attn_prefill_split_kernel, dim3(ntiles, Hq, NS),
                     dim3(DECODE_BLOCK), 0, current_stream(),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), bf16_cptr(q),
                     bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), tile_seq.data_ptr<int>(),
                     tile_pos0.data_ptr<int>(), (float)scale, Hq, Hkv, D, BS,
                     MAXB, GQ, NS);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(paged_attn_prefill_reduce_kernel, dim3(ntiles, Hq, 16),
                     dim3(DECODE_BLOCK), 0, current_stream(), bf16_ptr(out),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), Hq, D, NS);
  HIP_CHECK_KERNEL();
}

void paged_attn_prefill_mfma(torch::Tensor out, torch::Tensor q,
                             torch::Tensor kcache, torch::Tensor vcache,
                             torch::Tensor block_tables, torch::Tensor tile_q0,
                             torch::Tensor tile_qn, torch::Tensor tile_seq,
                             torch::Tensor tile_pos0, double scale) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  const int ntiles = tile_q0.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  TORCH_CHECK(D == 128, "MFMA prefill kernel requires head dim 128");
  TORCH_CHECK(Hq % Hkv == 0, "bad GQA ratio");
  if (ntiles == 0) return;
  hipLaunchKernelGGL(paged_attn_prefill_mfma_kernel, dim3(ntiles, Hq),
                     dim3(256), 0, current_stream(), bf16_ptr(out),
                     bf16_cptr(q), bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), tile_seq.data_ptr<int>(),
                     tile_pos0.data_ptr<int>(), (float)scale, Hq, Hkv, BS,
                     MAXB, GQ);
  HIP_CHECK_KERNEL();
}

void paged_attn_prefill_t12_split(
    torch::Tensor out, torch::Tensor q, torch::Tensor kcache,
    torch::Tensor vcache, torch::Tensor block_tables, torch::Tensor tile_q0,
    torch::Tensor tile_qn, torch::Tensor tile_seq, torch::Tensor tile_pos0,
    double scale, torch::Tensor part_m, torch::Tensor part_l,
    torch::Tensor part_acc) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  const int ntiles = tile_q0.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  const int NS8 = part_m.size(2);      // waves count as splits: NS8 = NS*8
  TORCH_CHECK(D == 128, "T12 split kernel requires head dim 128");
  TORCH_CHECK(Hq % Hkv == 0, "bad GQA ratio");
  TORCH_CHECK(NS8 % 8 == 0, "partial split dim must be a multiple of 8");
  TORCH_CHECK(part_m.size(0) >= ntiles && part_m.size(1) == Hq &&
              part_m.size(3) == QT, "workspace shape");
  if (ntiles == 0) return;
  hipLaunchKernelGGL(paged_attn_prefill_t12_split_kernel,
                     dim3(ntiles, Hq, NS8 / 8), dim3(512), 0,
                     current_stream(),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), bf16_cptr(q),
                     bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), tile_seq.data_ptr<int>(),
                     tile_pos0.data_ptr<int>(), (float)scale, Hq, Hkv, BS,
                     MAXB, GQ, NS8 / 8);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(paged_attn_prefill_reduce_kernel, dim3(ntiles, Hq, 16),
                     dim3(DECODE_BLOCK), 0, current_stream(), bf16_ptr(out),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), Hq, D, NS8);
  HIP_CHECK_KERNEL();
}

void paged_attn_prefill_mfma_split(
    torch::Tensor out, torch::Tensor q, torch::Tensor kcache,
    torch::Tensor vcache, torch::Tensor block_tables, torch::Tensor tile_q0,
    torch::Tensor tile_qn, torch::Tensor tile_seq, torch::Tensor tile_pos0,
    double scale, torch::Tensor part_m, torch::Tensor part_l,
    torch::Tensor part_acc) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  CHECK_GPU(part_m);
  CHECK_GPU(part_l);
  CHECK_GPU(part_acc);
  const int ntiles = tile_q0.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  const int NS = part_m.size(2);
  TORCH_CHECK(D == 128, "MFMA prefill kernel requires head dim 128");
  TORCH_CHECK(Hq % Hkv == 0, "bad GQA ratio");
  TORCH_CHECK(part_m.size(0) >= ntiles && part_m.size(1) == Hq &&
              part_m.size(3) == QT, "workspace shape");
  if (ntiles == 0) return;
  hipLaunchKernelGGL(paged_attn_prefill_mfma_split_kernel,
                     dim3(ntiles, Hq, NS), dim3(256), 0, current_stream(),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), bf16_cptr(q),
                     bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), tile_seq.data_ptr<int>(),
                     tile_pos0.data_ptr<int>(), (float)scale, Hq, Hkv, BS,
                     MAXB, GQ, NS);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(paged_attn_prefill_reduce_kernel, dim3(ntiles, Hq, 16),
                     dim3(DECODE_BLOCK), 0, current_stream(), bf16_ptr(out),
                     part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                     part_acc.data_ptr<float>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), Hq, D, NS);
  HIP_CHECK_KERNEL();
}

void cosine_sim_matrix(torch::Tensor out, torch::Tensor x) {
  CHECK_GPU(out);
  CHECK_GPU(x);
  TORCH_CHECK(x.scalar_type() == torch::kFloat32, "x must be f32");
  const int N = x.size(0);
  const int D = x.size(1);
  if (N == 0) return;
  hipLaunchKernelGGL(cosine_sim_kernel, dim3(N, N), dim3(256), 0,
                     current_stream(), out.data_ptr<float>(),
                     x.data_ptr<float>(), N, D);
  HIP_CHECK_KERNEL();
}

void gather_rows(torch::Tensor out, torch::Tensor src, torch::Tensor rows) {
  CHECK_GPU(out);
  CHECK_GPU(src);
  CHECK_GPU(rows);
  const int T = rows.size(0);
  const int n = src.size(1);
  TORCH_CHECK(n % 8 == 0, "row width must be a multiple of 8");
  if (T == 0) return;
  hipLaunchKernelGGL(gather_rows_kernel, dim3(T), dim3(256), 0,
                     current_stream(), bf16_ptr(out), bf16_cptr(src),
                     rows.data_ptr<int>(), n);
  HIP_CHECK_KERNEL();
}

void paged_attn_prefill_mfma32(torch::Tensor out, torch::Tensor q,
                               torch::Tensor kcache, torch::Tensor vcache,
                               torch::Tensor block_tables,
                               torch::Tensor tile_q0, torch::Tensor tile_qn,
                               torch::Tensor tile_seq,
                               torch::Tensor tile_pos0, double scale) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  const int ntiles = tile_q0.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  TORCH_CHECK(D == 128, "MFMA32 prefill kernel requires head dim 128");
  TORCH_CHECK(Hq % Hkv == 0, "bad GQA ratio");
  if (ntiles == 0) return;
  hipLaunchKernelGGL(paged_attn_prefill_mfma32_kernel, dim3(ntiles, Hq),
                     dim3(512), 0, current_stream(), bf16_ptr(out),
                     bf16_cptr(q), bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), tile_seq.data_ptr<int>(),
                     tile_pos0.data_ptr<int>(), (float)scale, Hq, Hkv, BS,
                     MAXB, GQ);
  HIP_CHECK_KERNEL();
}

void paged_attn_prefill_t12w(torch::Tensor out, torch::Tensor q,
                             torch::Tensor kcache, torch::Tensor vcache,
                             torch::Tensor block_tables,
                             torch::Tensor tile_q0, torch::Tensor tile_qn,
                             torch::Tensor tile_seq,
                             torch::Tensor tile_pos0, double scale) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  const int ntiles = tile_q0.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  TORCH_CHECK(D == 128, "T12W prefill kernel requires head dim 128");
  TORCH_CHECK(Hq % Hkv == 0, "bad GQA ratio");
  if (ntiles == 0) return;
  hipLaunchKernelGGL(paged_attn_prefill_t12w_kernel, dim3(ntiles, Hq),
                     dim3(256), 0, current_stream(), bf16_ptr(out),
                     bf16_cptr(q), bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), tile_seq.data_ptr<int>(),
                     tile_pos0.data_ptr<int>(), (float)scale, Hq, Hkv, BS,
                     MAXB, GQ);
  HIP_CHECK_KERNEL();
}

void paged_attn_prefill_t12(torch::Tensor out, torch::Tensor q,
                            torch::Tensor kcache, torch::Tensor vcache,
                            torch::Tensor block_tables,
                            torch::Tensor tile_q0, torch::Tensor tile_qn,
                            torch::Tensor tile_seq,
                            torch::Tensor tile_pos0, double scale) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  const int ntiles = tile_q0.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  TORCH_CHECK(D == 128, "T12 prefill kernel requires head dim 128");
  TORCH_CHECK(Hq % Hkv == 0, "bad GQA ratio");
  if (ntiles == 0) return;
  hipLaunchKernelGGL(paged_attn_prefill_t12_kernel, dim3(ntiles, Hq),
                     dim3(512), 0, current_stream(), bf16_ptr(out),
                     bf16_cptr(q), bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), tile_seq.data_ptr<int>(),
                     tile_pos0.data_ptr<int>(), (float)scale, Hq, Hkv, BS,
                     MAXB, GQ);
  HIP_CHECK_KERNEL();
}

void paged_attn_prefill_mfma64(torch::Tensor out, torch::Tensor q,
                               torch::Tensor kcache, torch::Tensor vcache,
                               torch::Tensor block_tables,
                               torch::Tensor tile_q0, torch::Tensor tile_qn,
                               torch::Tensor tile_seq,
                               torch::Tensor tile_pos0, double scale) {
  CHECK_GPU(out);
  CHECK_GPU(q);
  CHECK_GPU(kcache);
  CHECK_GPU(vcache);
  CHECK_GPU(block_tables);
  const int ntiles = tile_q0.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = kcache.size(1);
  const int BS = kcache.size(2);
  const int MAXB = block_tables.size(1);
  const int GQ = Hq / Hkv;
  TORCH_CHECK(D == 128, "MFMA64 prefill kernel requires head dim 128");
  TORCH_CHECK(Hq % Hkv == 0, "bad GQA ratio");
  if (ntiles == 0) return;
  hipLaunchKernelGGL(paged_attn_prefill_mfma64_kernel, dim3(ntiles, Hq),
                     dim3(512), 0, current_stream(), bf16_ptr(out),
                     bf16_cptr(q), bf16_cptr(kcache), bf16_cptr(vcache),
                     block_tables.data_ptr<int>(), tile_q0.data_ptr<int>(),
                     tile_qn.data_ptr<int>(), tile_seq.data_ptr<int>(),
                     tile_pos0.data_ptr<int>(), (float)scale, Hq, Hkv, BS,
                     MAXB, GQ);
  HIP_CHECK_KERNEL();
}

#include "forward.h"

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fused", &rmsnorm_fused,
        "Fused residual-add + RMSNorm (bf16, fp32 accum)");
  m.def("swiglu", &swiglu, "Fused SiLU(gate) * up");
  m.def("split_qkv", &split_qkv,
        "One-pass split of the fused QKV GEMM output into q/k/v");
  m.def("rope_inplace", &rope_inplace, "Rotate-half RoPE in place on q/k");
  m.def("kv_append", &kv_append, "Scatter K/V rows into the paged cache");
  m.def("paged_attn_decode", &paged_attn_decode,
        "Paged-KV GQA decode attention (one token per sequence)");
  m.def("paged_attn_decode_split2", &paged_attn_decode_split2,
        "EXPERIMENTAL: decode split v2 (256-token chunks, LDS block ids)");
  m.def("paged_attn_decode_split", &paged_attn_decode_split,
        "Flash-decoding: context-split decode attention + combine");
  m.def("paged_attn_decode_mfma", &paged_attn_decode_mfma,
        "MFMA flash-decode: context-split decode on the matrix cores "
        "(D=128; GQ heads padded into a 16-row tile) + combine");
  m.def("paged_attn_prefill", &paged_attn_prefill,
        "Paged-KV causal prefill attention over cached context");
  m.def("paged_attn_prefill_split", &paged_attn_prefill_split,
        "Context-split prefill attention + combine (small chunks)");
  m.def("paged_attn_prefill_mfma", &paged_attn_prefill_mfma,
        "MFMA-tiled prefill attention (D=128, matrix cores)");
  m.def("paged_attn_prefill_t12w", &paged_attn_prefill_t12w,
        "EXPERIMENTAL: T12 structure on the 32x32x16 MFMA (validate "
        "before use)");
  m.def("paged_attn_prefill_t12_split", &paged_attn_prefill_t12_split,
        "EXPERIMENTAL: context-split T12 prefill, wave-local chunks "
        "(partials over NS*8 splits) + combine");
  m.def("paged_attn_prefill_t12", &paged_attn_prefill_t12,
        "EXPERIMENTAL: 128-row T12 prefill (swapped QK^T, in-register "
        "softmax, permlane P exchange; validate before use)");
  m.def("paged_attn_prefill_mfma64", &paged_attn_prefill_mfma64,
        "EXPERIMENTAL: MFMA prefill, 64-row Q tiles (validate before use)");
  m.def("paged_attn_prefill_mfma32", &paged_attn_prefill_mfma32,
        "MFMA prefill attention, 32-row Q tiles (8 waves)");
  m.def("paged_attn_prefill_mfma_split", &paged_attn_prefill_mfma_split,
        "Context-split MFMA prefill (small chunks over long context)");
  m.def("cosine_sim_matrix", &cosine_sim_matrix,
        "Pairwise cosine similarity matrix (consensus vote)");
  m.def("gather_rows", &gather_rows, "Embedding row gather (bf16)");
  m.def("register_model", &qforward::register_model,
        "Register model weights with the C++ forward driver");
  m.def("unregister_model", &qforward::unregister_model,
        "Drop a registered model");
  m.def("llama_forward", &qforward::forward,
        "Full transformer forward (layer loop in native code)");
  m.def("llama_logits", &qforward::compute_logits,
        "lm_head logits for selected rows");
}
