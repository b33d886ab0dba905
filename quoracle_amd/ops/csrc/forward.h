// C++ forward driver: the whole per-layer loop in one native call.
//
// Eager (non-graph) engine steps — chunked prefill and mixed batches — were
// host-bound: ~400 Python op dispatches per forward cost ~15 ms while the
// GPU work is a few ms.  This driver holds the model weights in a C++
// registry and launches the entire transformer forward (GEMMs via
// at::matmul -> hipBLASLt, fused ops via the gfx950 kernels in this
// extension) from native code.  The Python LlamaModel registers itself once
// and delegates; the CPU/reference path stays in Python.
//
// Included by bindings.cpp in the same translation unit (after the kernel
// wrapper functions it calls).

#pragma once

#include <cstdlib>
#include <unordered_map>
#include <vector>

namespace qforward {

struct Layer {
  torch::Tensor attn_norm, wqkv, wo, ffn_norm;
  torch::Tensor w_gate_up, w_down;     // dense
  torch::Tensor router;                // MoE (undefined => dense)
};

struct Model {
  torch::Tensor embed, lm_head, final_norm;
  std::vector<Layer> layers;
  int64_t n_heads, n_kv_heads, head_dim, intermediate;
  double rope_theta, eps, scale;
  bool moe = false;
  int64_t top_k = 2;
};

inline std::unordered_map<std::string, Model> &registry() {
  static std::unordered_map<std::string, Model> models;
  return models;
}

inline void register_model(
    const std::string &key, torch::Tensor embed, torch::Tensor lm_head,
    torch::Tensor final_norm, const std::vector<std::vector<torch::Tensor>> &layers,
    int64_t n_heads, int64_t n_kv_heads, int64_t head_dim,
    int64_t intermediate, double rope_theta, double eps, bool moe,
    int64_t top_k) {
  Model m;
  m.embed = embed;
  m.lm_head = lm_head;
  m.final_norm = final_norm;
  m.n_heads = n_heads;
  m.n_kv_heads = n_kv_heads;
  m.head_dim = head_dim;
  m.intermediate = intermediate;
  m.rope_theta = rope_theta;
  m.eps = eps;
  m.scale = 1.0 / std::sqrt((double)head_dim);
  m.moe = moe;
  m.top_k = top_k;
  for (auto &lw : layers) {
    // dense: [attn_norm, wqkv, wo, ffn_norm, w_gate_up, w_down]
    // moe:   [attn_norm, wqkv, wo, ffn_norm, w_gate_up, w_down, router]
    Layer L;
    L.attn_norm = lw[0];
    L.wqkv = lw[1];
    L.wo = lw[2];
    L.ffn_norm = lw[3];
    L.w_gate_up = lw[4];
    L.w_down = lw[5];
    if (lw.size() > 6) L.router = lw[6];
    m.layers.push_back(std::move(L));
  }
  registry()[key] = std::move(m);
}

// Attention for one layer over the paged cache (decode rows + prefill tiles).
inline void attend(const Model &m, torch::Tensor attn_out, torch::Tensor q,
                   torch::Tensor kcache, torch::Tensor vcache,
                   torch::Tensor block_tables, int64_t n_decode,
                   c10::optional<torch::Tensor> ctx_lens, int64_t max_ctx,
                   c10::optional<torch::Tensor> tile_q0,
                   c10::optional<torch::Tensor> tile_qn,
                   c10::optional<torch::Tensor> tile_seq,
                   c10::optional<torch::Tensor> tile_pos0,
                   c10::optional<torch::Tensor> t32_q0,
                   c10::optional<torch::Tensor> t32_qn,
                   c10::optional<torch::Tensor> t32_seq,
                   c10::optional<torch::Tensor> t32_pos0, int64_t max_kv,
                   bool no_mfma) {
  const int64_t D = m.head_dim;
  if (n_decode > 0) {
    auto out_d = attn_out.narrow(0, 0, n_decode);
    auto q_d = q.narrow(0, 0, n_decode).contiguous();
    if (max_ctx >= 1024) {
      const int64_t B = n_decode, Hq = m.n_heads;
      int64_t ns = std::min<int64_t>(32, std::max<int64_t>(2, max_ctx / 256));
      auto opts = torch::TensorOptions()
                      .dtype(torch::kFloat32).device(q.device());
      auto pm = torch::empty({B, Hq, ns}, opts);
      auto pl = torch::empty({B, Hq, ns}, opts);
      auto pa = torch::empty({B, Hq, ns, D}, opts);
      static const bool dec_valu =
          std::getenv("QUORACLE_DECODE_VALU") != nullptr;
      if (D == 128 && !dec_valu)
        paged_attn_decode_mfma(out_d, q_d, kcache, vcache, block_tables,
                               ctx_lens.value(), m.scale, pm, pl, pa);
      else
        paged_attn_decode_split(out_d, q_d, kcache, vcache, block_tables,
                                ctx_lens.value(), m.scale, pm, pl, pa);
    } else {
      paged_attn_decode(out_d, q_d, kcache, vcache, block_tables,
                        ctx_lens.value(), m.scale);
    }
    // narrow(0,..) of attn_out is a view -> writes land in attn_out; but
    // q narrow+contiguous copied; out_d IS a view (contiguous prefix) so ok
  }
  if (tile_q0.has_value() && tile_q0->numel() > 0) {
    const int64_t ntiles = tile_q0->size(0), Hq = m.n_heads;
    int64_t ns = std::max<int64_t>(
        1, std::min<int64_t>(48, 4096 / std::max<int64_t>(1, ntiles * Hq)));
    auto opts = torch::TensorOptions()
                    .dtype(torch::kFloat32).device(q.device());
    if (D == 128 && !no_mfma) {
      if (ns > 1 && max_kv >= 1024) {
        auto pm = torch::empty({ntiles, Hq, ns, 16}, opts);
        auto pl = torch::empty({ntiles, Hq, ns, 16}, opts);
        auto pa = torch::empty({ntiles, Hq, ns, 16, D}, opts);
        paged_attn_prefill_mfma_split(attn_out, q, kcache, vcache,
                                      block_tables, *tile_q0, *tile_qn,
                                      *tile_seq, *tile_pos0, m.scale, pm, pl,
                                      pa);
      } else if (t32_q0.has_value() && t32_q0->numel() > 0) {
        // big prefill: T12 128-row tiles are the default (352 TF measured
        // vs 267 for mfma64 / 169 for mfma32 on the bench shape, r2);
        // QUORACLE_MFMA64 / QUORACLE_MFMA32 fall back (the engine sizes
        // the tiles to match)
        static const bool use32 = std::getenv("QUORACLE_MFMA32") != nullptr;
        static const bool use64 = std::getenv("QUORACLE_MFMA64") != nullptr;
        if (use32)
          paged_attn_prefill_mfma32(attn_out, q, kcache, vcache,
                                    block_tables, *t32_q0, *t32_qn,
                                    *t32_seq, *t32_pos0, m.scale);
        else if (use64)
          paged_attn_prefill_mfma64(attn_out, q, kcache, vcache,
                                    block_tables, *t32_q0, *t32_qn,
                                    *t32_seq, *t32_pos0, m.scale);
        else
          paged_attn_prefill_t12(attn_out, q, kcache, vcache,
                                 block_tables, *t32_q0, *t32_qn,
                                 *t32_seq, *t32_pos0, m.scale);
      } else {
        paged_attn_prefill_mfma(attn_out, q, kcache, vcache, block_tables,
                                *tile_q0, *tile_qn, *tile_seq, *tile_pos0,
                                m.scale);
      }
    } else if (ns > 1 && max_kv >= 1024) {
      auto pm = torch::empty({ntiles, Hq, ns, 16}, opts);
      auto pl = torch::empty({ntiles, Hq, ns, 16}, opts);
      auto pa = torch::empty({ntiles, Hq, ns, 16, D}, opts);
      paged_attn_prefill_split(attn_out, q, kcache, vcache, block_tables,
                               *tile_q0, *tile_qn, *tile_seq, *tile_pos0,
                               m.scale, pm, pl, pa);
    } else {
      paged_attn_prefill(attn_out, q, kcache, vcache, block_tables, *tile_q0,
                         *tile_qn, *tile_seq, *tile_pos0, m.scale);
    }
  }
}

// Rows at or below this take the capacity-padded batched path (zero host
// syncs, hipGraph-capturable).  Above it (big prefill chunks) the GEMMs are
// compute-bound and the 4x padding waste costs more than one counts sync.
constexpr int64_t MOE_BMM_MAX_ROWS = 512;

// Capacity-padded batched MoE FFN: every expert gets a fixed bin of T rows
// (an expert can receive at most one row per token), tokens are scattered
// into their expert bin on the GPU, and both FFN GEMMs run as ONE bmm over
// [E, T, *].  No counts.cpu() — decode steps stay sync-free and capture
// into hipGraphs.  Padding is free here: at decode sizes the GEMMs are
// bound by streaming the expert weights, not by rows.
inline torch::Tensor moe_ffn_bmm(const Model &m, const Layer &L,
                                 torch::Tensor h, torch::Tensor weights,
                                 torch::Tensor experts) {
  const int64_t T = h.size(0);
  const int64_t K = m.top_k;
  const int64_t E = L.router.size(1);
  const int64_t S = T * K, C = T;
  const int64_t hidden = h.size(1);
  auto expert_flat = experts.reshape({-1});
  auto order = expert_flat.argsort(/*stable=*/true);
  auto e_sorted = expert_flat.index_select(0, order);
  auto token_of = order.div(K, "floor");
  // counts via scatter_add, NOT at::bincount: bincount computes its
  // output size with a device->host sync, which is illegal inside
  // hipGraph capture (this was the dual-preset capture failure)
  auto counts = torch::zeros({E}, expert_flat.options());
  counts.scatter_add_(0, expert_flat, torch::ones_like(expert_flat));
  auto raw_off = at::cumsum(counts, 0) - counts;   // exclusive prefix
  auto pos = at::arange(S, order.options()) -
             raw_off.index_select(0, e_sorted);
  auto idx = e_sorted * C + pos;                   // slot in the bins
  auto bins = torch::zeros({E * C, hidden}, h.options());
  bins.index_copy_(0, idx, h.index_select(0, token_of));
  auto gu = at::bmm(bins.view({E, C, hidden}), L.w_gate_up);
  auto act = torch::empty({E * C, m.intermediate}, h.options());
  swiglu(act, gu.view({E * C, gu.size(2)}));
  auto down = at::bmm(act.view({E, C, m.intermediate}), L.w_down);
  auto w_sorted = weights.reshape({-1}).index_select(0, order).unsqueeze(1);
  auto vals = down.view({E * C, hidden}).index_select(0, idx) * w_sorted;
  auto out = torch::zeros_like(h);
  out.index_add_(0, token_of, vals);
  return out;
}

inline torch::Tensor moe_ffn(const Model &m, const Layer &L,
                             torch::Tensor h) {
  auto logits = at::matmul(h, L.router).to(torch::kFloat32);
  auto probs = at::softmax(logits, -1);
  auto topk = probs.topk(m.top_k, -1);
  auto weights = std::get<0>(topk);
  auto experts = std::get<1>(topk);
  weights = (weights / weights.sum(-1, true)).to(h.scalar_type());
  if (h.size(0) * m.top_k <= MOE_BMM_MAX_ROWS && h.is_cuda())
    return moe_ffn_bmm(m, L, h, weights, experts);
  // big prefill chunks: token-sorted dispatch, one contiguous GEMM slice
  // per expert (compute-bound — the single counts sync amortizes over the
  // whole chunk), single weighted scatter-add
  const int64_t E = L.router.size(1);
  auto expert_flat = experts.reshape({-1});
  auto order = expert_flat.argsort(/*stable=*/true);
  auto token_of = order.div(m.top_k, "floor");
  auto h_sorted = h.index_select(0, token_of);
  auto counts =
      at::bincount(expert_flat, /*weights=*/{}, /*minlength=*/E).cpu();
  auto counts_a = counts.accessor<int64_t, 1>();
  auto down_sorted = torch::empty_like(h_sorted);
  int64_t start = 0;
  for (int64_t e = 0; e < E; ++e) {
    const int64_t n = counts_a[e];
    if (n == 0) continue;
    auto he = h_sorted.narrow(0, start, n);
    auto gu = at::matmul(he, L.w_gate_up[e]);
    auto act = torch::empty({n, m.intermediate}, h.options());
    swiglu(act, gu);
    down_sorted.narrow(0, start, n) = at::matmul(act, L.w_down[e]);
    start += n;
  }
  auto w_sorted =
      weights.reshape({-1}).index_select(0, order).unsqueeze(1);
  auto out = torch::zeros_like(h);
  out.index_add_(0, token_of, down_sorted * w_sorted);
  return out;
}

inline torch::Tensor forward(
    const std::string &key, torch::Tensor tokens, torch::Tensor positions,
    torch::Tensor slots, torch::Tensor block_tables, int64_t n_decode,
    c10::optional<torch::Tensor> ctx_lens, int64_t max_ctx,
    c10::optional<torch::Tensor> tile_q0, c10::optional<torch::Tensor> tile_qn,
    c10::optional<torch::Tensor> tile_seq,
    c10::optional<torch::Tensor> tile_pos0,
    c10::optional<torch::Tensor> t32_q0, c10::optional<torch::Tensor> t32_qn,
    c10::optional<torch::Tensor> t32_seq,
    c10::optional<torch::Tensor> t32_pos0, int64_t max_kv,
    std::vector<torch::Tensor> kcaches, std::vector<torch::Tensor> vcaches,
    bool no_mfma) {
  auto it = registry().find(key);
  TORCH_CHECK(it != registry().end(), "model not registered: ", key);
  const Model &m = it->second;
  const int64_t T = tokens.size(0);
  const int64_t hidden = m.embed.size(1);
  const int64_t q_dim = m.n_heads * m.head_dim;
  const int64_t kv_dim = m.n_kv_heads * m.head_dim;

  auto res = torch::empty({T, hidden}, m.embed.options());
  gather_rows(res, m.embed, tokens);
  auto h = torch::empty_like(res);
  rmsnorm_fused(h, res, c10::nullopt, m.layers[0].attn_norm, m.eps);

  auto attn_out = torch::empty({T, m.n_heads, m.head_dim}, m.embed.options());

  auto q = torch::empty({T, m.n_heads, m.head_dim}, m.embed.options());
  auto k = torch::empty({T, m.n_kv_heads, m.head_dim}, m.embed.options());
  auto v = torch::empty({T, m.n_kv_heads, m.head_dim}, m.embed.options());
  for (size_t li = 0; li < m.layers.size(); ++li) {
    const Layer &L = m.layers[li];
    auto qkv = at::matmul(h, L.wqkv);
    // one fused pass instead of three narrow().contiguous() copy chains
    split_qkv(q, k, v, qkv);
    rope_inplace(q, k, positions, m.rope_theta);
    kv_append(kcaches[li], vcaches[li], k, v, slots);
    attend(m, attn_out, q, kcaches[li], vcaches[li], block_tables, n_decode,
           ctx_lens, max_ctx, tile_q0, tile_qn, tile_seq, tile_pos0,
           t32_q0, t32_qn, t32_seq, t32_pos0, max_kv, no_mfma);
    auto proj = at::matmul(attn_out.view({T, q_dim}), L.wo);
    rmsnorm_fused(h, proj, res, L.ffn_norm, m.eps);
    torch::Tensor ffn;
    if (m.moe) {
      ffn = moe_ffn(m, L, h);
    } else {
      auto gu = at::matmul(h, L.w_gate_up);
      auto act = torch::empty({T, m.intermediate}, h.options());
      swiglu(act, gu);
      ffn = at::matmul(act, L.w_down);
    }
    auto &next_norm = (li + 1 < m.layers.size())
                          ? m.layers[li + 1].attn_norm : m.final_norm;
    rmsnorm_fused(h, ffn, res, next_norm, m.eps);
  }
  return h;
}

inline torch::Tensor compute_logits(const std::string &key,
                                    torch::Tensor hidden,
                                    torch::Tensor rows) {
  auto it = registry().find(key);
  TORCH_CHECK(it != registry().end(), "model not registered: ", key);
  return at::matmul(hidden.index_select(0, rows), it->second.lm_head.t())
      .to(torch::kFloat32);
}

inline void unregister_model(const std::string &key) {
  registry().erase(key);
}

}  // namespace qforward
