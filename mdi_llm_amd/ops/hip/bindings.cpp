// Torch bindings for the hand-written CDNA4 decode kernels.
// Host-only TU: tensor checks + pointer extraction + launcher calls on the
// current torch HIP stream.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

#include "decode_kernels.h"

namespace {

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

inline void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

inline void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

inline void check_i32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kInt32, name, " must be int32");
}

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
             double eps) {
  check_bf16(out, "out");
  check_bf16(x, "x");
  check_bf16(w, "w");
  const int n = (int)w.numel();
  const int nb = (int)(x.numel() / n);
  TORCH_CHECK(n % 8 == 0, "n must be a multiple of 8");
  TORCH_CHECK(x.numel() == (int64_t)nb * n && out.numel() == x.numel(),
              "size mismatch");
  launch_rmsnorm(out.data_ptr(), x.data_ptr(), w.data_ptr(), n, (float)eps,
                 nb, cur_stream());
}

void layernorm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
               c10::optional<torch::Tensor> b, double eps) {
  check_bf16(out, "out");
  check_bf16(x, "x");
  check_bf16(w, "w");
  const int n = (int)x.numel();
  TORCH_CHECK(n % 8 == 0, "n must be a multiple of 8");
  const void* bp = nullptr;
  if (b.has_value()) {
    check_bf16(*b, "b");
    bp = b->data_ptr();
  }
  launch_layernorm(out.data_ptr(), x.data_ptr(), w.data_ptr(), bp, n,
                   (float)eps, cur_stream());
}

void gemv(torch::Tensor out, torch::Tensor W, torch::Tensor x,
          c10::optional<torch::Tensor> bias, c10::optional<torch::Tensor> res,
          int64_t epilogue, c10::optional<torch::Tensor> norm_w,
          c10::optional<torch::Tensor> norm_b, int64_t norm_kind,
          double eps, int64_t rows, c10::optional<torch::Tensor> eidx,
          int64_t estride) {
  check_bf16(out, "out");
  check_bf16(W, "W");
  check_bf16(x, "x");
  const int K = (int)x.numel();
  const int M = (int)out.numel();
  const int* eip = nullptr;
  if (eidx.has_value()) {
    check_i32(*eidx, "eidx");
    TORCH_CHECK(norm_kind != 2, "expert indirection: no LayerNorm");
    TORCH_CHECK(W.numel() % ((int64_t)M * K) == 0, "stacked W shape");
    eip = eidx->data_ptr<int>();
  } else {
    TORCH_CHECK(W.numel() == (int64_t)M * K, "W shape mismatch");
  }
  TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8");
  const void* bp = nullptr;
  const void* rp = nullptr;
  const void* nwp = nullptr;
  const void* nbp = nullptr;
  if (bias.has_value()) {
    check_bf16(*bias, "bias");
    TORCH_CHECK(bias->numel() == M, "bias size");
    bp = bias->data_ptr();
  }
  if (res.has_value()) {
    check_bf16(*res, "res");
    TORCH_CHECK(res->numel() == M, "res size");
    rp = res->data_ptr();
  }
  if (norm_kind != 0) {
    TORCH_CHECK(norm_w.has_value(), "norm_w required with norm_kind");
    check_bf16(*norm_w, "norm_w");
    nwp = norm_w->data_ptr();
    if (norm_b.has_value()) {
      check_bf16(*norm_b, "norm_b");
      nbp = norm_b->data_ptr();
    }
  }
  launch_gemv(out.data_ptr(), W.data_ptr(), x.data_ptr(), bp, rp, nwp, nbp,
              (float)eps, M, K, (int)epilogue, (int)norm_kind, (int)rows,
              eip, (long long)estride, cur_stream());
}

void gemv_fp8(torch::Tensor out, torch::Tensor W, torch::Tensor wscale,
              torch::Tensor x, c10::optional<torch::Tensor> bias,
              c10::optional<torch::Tensor> res, int64_t epilogue,
              c10::optional<torch::Tensor> norm_w,
              c10::optional<torch::Tensor> norm_b, int64_t norm_kind,
              double eps, int64_t rows) {
  check_bf16(out, "out");
  check_bf16(x, "x");
  check_f32(wscale, "wscale");
  TORCH_CHECK(W.is_cuda() && W.element_size() == 1 && W.is_contiguous(),
              "W must be contiguous fp8/uint8 on GPU");
  const int K = (int)x.numel();
  const int M = (int)out.numel();
  TORCH_CHECK(W.numel() == (int64_t)M * K, "W shape mismatch");
  TORCH_CHECK(K % 16 == 0, "K must be a multiple of 16 for fp8 weights");
  TORCH_CHECK(wscale.numel() == M, "wscale size");
  const void* bp = nullptr;
  const void* rp = nullptr;
  const void* nwp = nullptr;
  const void* nbp = nullptr;
  if (bias.has_value()) { check_bf16(*bias, "bias"); bp = bias->data_ptr(); }
  if (res.has_value()) { check_bf16(*res, "res"); rp = res->data_ptr(); }
  if (norm_kind != 0) {
    check_bf16(*norm_w, "norm_w");
    nwp = norm_w->data_ptr();
    if (norm_b.has_value()) { check_bf16(*norm_b, "norm_b"); nbp = norm_b->data_ptr(); }
  }
  launch_gemv_fp8(out.data_ptr(), W.data_ptr(), wscale.data_ptr<float>(),
                  x.data_ptr(), bp, rp, nwp, nbp, (float)eps, M, K,
                  (int)epilogue, (int)norm_kind, (int)rows, cur_stream());
}

void gemv_swiglu_fp8(torch::Tensor out, torch::Tensor Wg,
                     torch::Tensor gscale, torch::Tensor Wu,
                     torch::Tensor uscale, torch::Tensor x, bool gelu_gate,
                     c10::optional<torch::Tensor> norm_w,
                     c10::optional<torch::Tensor> norm_b, int64_t norm_kind,
                     double eps) {
  check_bf16(out, "out");
  check_bf16(x, "x");
  check_f32(gscale, "gscale");
  check_f32(uscale, "uscale");
  TORCH_CHECK(Wg.element_size() == 1 && Wu.element_size() == 1,
              "fp8 weights must be 1-byte");
  const int K = (int)x.numel();
  const int M = (int)out.numel();
  TORCH_CHECK(K % 16 == 0, "K must be a multiple of 16 for fp8 weights");
  const void* nwp = nullptr;
  const void* nbp = nullptr;
  if (norm_kind != 0) {
    check_bf16(*norm_w, "norm_w");
    nwp = norm_w->data_ptr();
    if (norm_b.has_value()) { check_bf16(*norm_b, "norm_b"); nbp = norm_b->data_ptr(); }
  }
  launch_gemv_swiglu_fp8(out.data_ptr(), Wg.data_ptr(),
                         gscale.data_ptr<float>(), Wu.data_ptr(),
                         uscale.data_ptr<float>(), x.data_ptr(), nwp, nbp,
                         (float)eps, M, K, gelu_gate ? 1 : 0,
                         (int)norm_kind, cur_stream());
}

void gemv_swiglu(torch::Tensor out, torch::Tensor Wg, torch::Tensor Wu,
                 torch::Tensor x, bool gelu_gate,
                 c10::optional<torch::Tensor> norm_w,
                 c10::optional<torch::Tensor> norm_b, int64_t norm_kind,
                 double eps, c10::optional<torch::Tensor> eidx,
                 int64_t estride, c10::optional<torch::Tensor> escale) {
  check_bf16(out, "out");
  check_bf16(Wg, "Wg");
  check_bf16(Wu, "Wu");
  check_bf16(x, "x");
  const int K = (int)x.numel();
  const int M = (int)out.numel();
  const int* eip = nullptr;
  const float* esp = nullptr;
  if (eidx.has_value()) {
    check_i32(*eidx, "eidx");
    TORCH_CHECK(Wg.numel() % ((int64_t)M * K) == 0 &&
                    Wu.numel() % ((int64_t)M * K) == 0,
                "stacked weight shape");
    eip = eidx->data_ptr<int>();
  } else {
    TORCH_CHECK(Wg.numel() == (int64_t)M * K && Wu.numel() == (int64_t)M * K,
                "weight shape mismatch");
  }
  if (escale.has_value()) {
    check_f32(*escale, "escale");
    esp = escale->data_ptr<float>();
  }
  TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8");
  const void* nwp = nullptr;
  const void* nbp = nullptr;
  if (norm_kind != 0) {
    TORCH_CHECK(norm_w.has_value(), "norm_w required with norm_kind");
    check_bf16(*norm_w, "norm_w");
    nwp = norm_w->data_ptr();
    if (norm_b.has_value()) {
      check_bf16(*norm_b, "norm_b");
      nbp = norm_b->data_ptr();
    }
  }
  launch_gemv_swiglu(out.data_ptr(), Wg.data_ptr(), Wu.data_ptr(),
                     x.data_ptr(), nwp, nbp, (float)eps, M, K,
                     gelu_gate ? 1 : 0, (int)norm_kind, eip,
                     (long long)estride, esp, cur_stream());
}

void swiglu_mul(torch::Tensor out, torch::Tensor g, torch::Tensor u,
                bool gelu_gate) {
  check_bf16(out, "out");
  check_bf16(g, "g");
  check_bf16(u, "u");
  TORCH_CHECK(out.numel() == g.numel() && g.numel() == u.numel() &&
                  g.numel() % 8 == 0,
              "swiglu_mul: equal sizes, multiple of 8");
  launch_swiglu_mul(out.data_ptr(), g.data_ptr(), u.data_ptr(),
                    (long long)g.numel(), gelu_gate ? 1 : 0, cur_stream());
}

void moe_gate_topk(torch::Tensor eidx, torch::Tensor escale,
                   torch::Tensor logits, int64_t k) {
  check_i32(eidx, "eidx");
  check_f32(escale, "escale");
  check_bf16(logits, "logits");
  const int n_e = (int)logits.numel();
  TORCH_CHECK(n_e <= 64 && k <= 8 && k >= 1 && eidx.numel() >= k &&
                  escale.numel() >= k,
              "moe_gate_topk: n_e <= 64, 1 <= k <= 8");
  launch_moe_gate_topk(eidx.data_ptr<int>(), escale.data_ptr<float>(),
                       logits.data_ptr(), n_e, (int)k, cur_stream());
}

void embed(torch::Tensor out, torch::Tensor wte, torch::Tensor token,
           double scale) {
  check_bf16(out, "out");
  check_bf16(wte, "wte");
  check_i32(token, "token");
  const int n = (int)out.numel();
  TORCH_CHECK(n % 8 == 0, "n_embd must be a multiple of 8");
  launch_embed(out.data_ptr(), wte.data_ptr(), token.data_ptr<int>(), n,
               (float)scale, cur_stream());
}

void rope_kv_append(torch::Tensor qkv, torch::Tensor kpool,
                    torch::Tensor vpool, torch::Tensor cos_t,
                    torch::Tensor sin_t, torch::Tensor pos,
                    torch::Tensor slot, int64_t layer) {
  check_bf16(qkv, "qkv");
  check_bf16(kpool, "kpool");
  check_bf16(vpool, "vpool");
  check_f32(cos_t, "cos");
  check_f32(sin_t, "sin");
  check_i32(pos, "pos");
  check_i32(slot, "slot");
  // pool: [slots, layers, kv_heads, max_seq, head_size]
  TORCH_CHECK(kpool.dim() == 5, "kpool must be 5-D");
  const int n_layers_pool = (int)kpool.size(1);
  const int n_kv = (int)kpool.size(2);
  const int max_seq = (int)kpool.size(3);
  const int hs = (int)kpool.size(4);
  const int rope_n_elem = (int)cos_t.size(1);
  const int qkv_dim = (int)qkv.numel();
  const int qpk = qkv_dim / (n_kv * hs) - 2;
  TORCH_CHECK(qpk >= 1, "bad qkv length");
  launch_rope_kv_append(qkv.data_ptr(), kpool.data_ptr(), vpool.data_ptr(),
                        cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                        pos.data_ptr<int>(), slot.data_ptr<int>(),
                        (int)layer, n_layers_pool, n_kv, max_seq, hs,
                        rope_n_elem, qpk, cur_stream());
}

void attn_decode(torch::Tensor out, torch::Tensor part_o,
                 torch::Tensor part_ml, torch::Tensor qkv,
                 torch::Tensor kpool, torch::Tensor vpool,
                 torch::Tensor cos_t, torch::Tensor sin_t, torch::Tensor pos,
                 torch::Tensor slot, int64_t layer, int64_t n_chunks,
                 double scale, int64_t n_batch,
                 c10::optional<torch::Tensor> kscale,
                 c10::optional<torch::Tensor> vscale,
                 int64_t force_split) {
  check_bf16(out, "out");
  check_f32(part_o, "part_o");
  check_f32(part_ml, "part_ml");
  check_bf16(qkv, "qkv");
  const bool kv8 = kscale.has_value();
  if (kv8) {
    TORCH_CHECK(kpool.scalar_type() == torch::kUInt8 &&
                    vpool.scalar_type() == torch::kUInt8,
                "fp8 KV cache must be uint8 pools");
    check_f32(*kscale, "kscale");
    check_f32(*vscale, "vscale");
  } else {
    check_bf16(kpool, "kpool");
    check_bf16(vpool, "vpool");
  }
  check_f32(cos_t, "cos");
  check_f32(sin_t, "sin");
  check_i32(pos, "pos");
  check_i32(slot, "slot");
  const int n_layers_pool = (int)kpool.size(1);
  const int n_kv = (int)kpool.size(2);
  const int max_seq = (int)kpool.size(3);
  const int hs = (int)kpool.size(4);
  const int rope_ne = cos_t.dim() > 1 ? (int)cos_t.size(1) : 0;
  const int nb = (int)n_batch;
  const int beff = nb > 0 ? nb : 1;
  const int qkv_dim = (int)(qkv.numel() / beff);
  const int qpk = qkv_dim / (n_kv * hs) - 2;
  const int n_head = n_kv * qpk;
  if (nb > 0) {
    TORCH_CHECK(pos.numel() >= nb && slot.numel() >= nb,
                "pos/slot arrays too small for batch");
  }
  TORCH_CHECK(part_o.numel() >= (int64_t)beff * n_head * n_chunks * hs,
              "part_o too small");
  TORCH_CHECK(part_ml.numel() >= (int64_t)beff * n_head * n_chunks * 2,
              "part_ml too small");
  TORCH_CHECK(out.numel() == (int64_t)beff * n_head * hs, "out size");
  int rc = launch_attn_decode(
      out.data_ptr(), part_o.data_ptr<float>(), part_ml.data_ptr<float>(),
      qkv.data_ptr(), kpool.data_ptr(), vpool.data_ptr(),
      kv8 ? kscale->data_ptr<float>() : nullptr,
      kv8 ? vscale->data_ptr<float>() : nullptr,
      rope_ne ? cos_t.data_ptr<float>() : nullptr,
      rope_ne ? sin_t.data_ptr<float>() : nullptr, rope_ne,
      pos.data_ptr<int>(), slot.data_ptr<int>(), (int)layer, n_layers_pool,
      n_kv, max_seq, hs, qpk, (int)n_chunks, (float)scale, nb,
      (int)force_split, cur_stream());
  TORCH_CHECK(rc == 0, "attn_decode: unsupported geometry qpk=", qpk,
              " head_size=", hs);
}

void attn_proj(torch::Tensor out, torch::Tensor qkv, torch::Tensor kpool,
               torch::Tensor vpool, torch::Tensor cos_t,
               torch::Tensor sin_t, torch::Tensor pos, torch::Tensor slot,
               int64_t layer, double scale, torch::Tensor W,
               c10::optional<torch::Tensor> bias,
               c10::optional<torch::Tensor> res, torch::Tensor gran) {
  check_bf16(out, "out");
  check_bf16(qkv, "qkv");
  check_bf16(kpool, "kpool");
  check_bf16(vpool, "vpool");
  check_f32(cos_t, "cos");
  check_f32(sin_t, "sin");
  check_i32(pos, "pos");
  check_i32(slot, "slot");
  check_bf16(W, "W");
  TORCH_CHECK(gran.is_cuda() && gran.scalar_type() == torch::kInt32,
              "gran must be int32 on GPU");
  const int n_layers_pool = (int)kpool.size(1);
  const int n_kv = (int)kpool.size(2);
  const int max_seq = (int)kpool.size(3);
  const int hs = (int)kpool.size(4);
  const int rope_ne = cos_t.dim() > 1 ? (int)cos_t.size(1) : 0;
  const int qkv_dim = (int)qkv.numel();
  const int qpk = qkv_dim / (n_kv * hs) - 2;
  const int K = n_kv * qpk * hs;
  const int M = (int)W.size(0);
  TORCH_CHECK((int)W.size(1) == K, "proj W inner dim mismatch");
  TORCH_CHECK(out.numel() >= M, "out too small");
  TORCH_CHECK(gran.numel() >= K / 2 + 16, "y/flag scratch too small");
  TORCH_CHECK(K % 128 == 0 && K / 2 <= 4096,
              "attn_proj: unsupported K for granule sweep");
  int rc = launch_attn_proj(
      out.data_ptr(), qkv.data_ptr(), kpool.data_ptr(), vpool.data_ptr(),
      rope_ne ? cos_t.data_ptr<float>() : nullptr,
      rope_ne ? sin_t.data_ptr<float>() : nullptr, rope_ne,
      pos.data_ptr<int>(), slot.data_ptr<int>(), (int)layer, n_layers_pool,
      n_kv, max_seq, hs, qpk, (float)scale, W.data_ptr(),
      bias.has_value() ? bias->data_ptr() : nullptr,
      res.has_value() ? res->data_ptr() : nullptr, gran.data_ptr(), M,
      cur_stream());
  TORCH_CHECK(rc == 0, "attn_proj: unsupported geometry qpk=", qpk,
              " head_size=", hs, " max_seq=", max_seq);
}

void sample(torch::Tensor out_token, torch::Tensor logits,
            torch::Tensor scratch, double temperature, int64_t top_k,
            bool noise, int64_t seed, c10::optional<torch::Tensor> pos,
            c10::optional<torch::Tensor> slot, int64_t n_batch,
            double top_p, c10::optional<torch::Tensor> token_table,
            c10::optional<torch::Tensor> pos_table,
            c10::optional<torch::Tensor> adv_slot, int64_t adv_pos,
            int64_t pos_bias) {
  check_i32(out_token, "out_token");
  check_bf16(logits, "logits");
  const int nb = n_batch > 0 ? (int)n_batch : 1;
  TORCH_CHECK(scratch.is_cuda() && scratch.scalar_type() == torch::kInt32 &&
                  scratch.numel() >= 520 * nb,
              "scratch must be >=520 int32 per sample on GPU");
  TORCH_CHECK(out_token.numel() >= nb, "out_token too small");
  const int* pp = nullptr;
  const int* sp = nullptr;
  if (pos.has_value()) {
    check_i32(*pos, "pos");
    TORCH_CHECK(pos->numel() >= nb, "pos too small");
    pp = pos->data_ptr<int>();
  }
  if (slot.has_value()) {
    check_i32(*slot, "slot");
    TORCH_CHECK(slot->numel() >= nb, "slot too small");
    sp = slot->data_ptr<int>();
  }
  int* ttp = nullptr;
  int* ptp = nullptr;
  const int* asp = nullptr;
  if (token_table.has_value()) {
    check_i32(*token_table, "token_table");
    ttp = token_table->data_ptr<int>();
  }
  if (pos_table.has_value()) {
    check_i32(*pos_table, "pos_table");
    ptp = pos_table->data_ptr<int>();
  }
  if (adv_slot.has_value()) {
    check_i32(*adv_slot, "adv_slot");
    asp = adv_slot->data_ptr<int>();
  }
  const int V = (int)(logits.numel() / nb);
  launch_sample(out_token.data_ptr(), logits.data_ptr(), V,
                scratch.data_ptr(), (float)temperature, (int)top_k,
                (float)top_p, noise ? 1 : 0, (unsigned)(int64_t)seed, pp, sp,
                (int)n_batch, ttp, ptp, asp, (int)adv_pos, (int)pos_bias,
                cur_stream());
}

void mtile_gemm(torch::Tensor Y, torch::Tensor W, torch::Tensor X,
                c10::optional<torch::Tensor> bias,
                c10::optional<torch::Tensor> res) {
  check_bf16(Y, "Y");
  check_bf16(W, "W");
  check_bf16(X, "X");
  const int Bsz = (int)X.size(0);
  const int K = (int)X.size(1);
  const int M = (int)W.size(0);
  TORCH_CHECK((int)W.size(1) == K, "mtile_gemm: K mismatch");
  TORCH_CHECK(Y.numel() >= (int64_t)Bsz * M, "mtile_gemm: Y too small");
  int rc = launch_mtile_gemm(
      Y.data_ptr(), W.data_ptr(), X.data_ptr(),
      bias.has_value() ? bias->data_ptr() : nullptr,
      res.has_value() ? res->data_ptr() : nullptr, Bsz, M, K,
      cur_stream());
  TORCH_CHECK(rc == 0, "mtile_gemm: unsupported shape B=", Bsz, " K=", K,
              " M=", M);
}

void route_env(torch::Tensor hdr, torch::Tensor slot_out,
               torch::Tensor pos_table, int64_t dummy_slot) {
  check_i32(hdr, "hdr");
  check_i32(slot_out, "slot_out");
  check_i32(pos_table, "pos_table");
  launch_route_env(hdr.data_ptr<int>(), slot_out.data_ptr<int>(),
                   pos_table.data_ptr<int>(), (int)dummy_slot, cur_stream());
}

void stage_slot(torch::Tensor slot, c10::optional<torch::Tensor> pos_out,
                c10::optional<torch::Tensor> token_out,
                c10::optional<torch::Tensor> pos_table,
                c10::optional<torch::Tensor> token_table,
                c10::optional<torch::Tensor> pos_table_mut,
                int64_t adv_pos) {
  check_i32(slot, "slot");
  int* po = pos_out.has_value() ? pos_out->data_ptr<int>() : nullptr;
  int* to = token_out.has_value() ? token_out->data_ptr<int>() : nullptr;
  const int* pt =
      pos_table.has_value() ? pos_table->data_ptr<int>() : nullptr;
  const int* tt =
      token_table.has_value() ? token_table->data_ptr<int>() : nullptr;
  int* ptm =
      pos_table_mut.has_value() ? pos_table_mut->data_ptr<int>() : nullptr;
  launch_stage_slot(po, to, pt, tt, ptm, slot.data_ptr<int>(),
                    (int)adv_pos, cur_stream());
}

void rope_prefill_append(torch::Tensor qkv, torch::Tensor kpool,
                         torch::Tensor vpool, torch::Tensor cos_t,
                         torch::Tensor sin_t, int64_t pos0, int64_t slot,
                         int64_t layer,
                         c10::optional<torch::Tensor> kscale,
                         c10::optional<torch::Tensor> vscale) {
  check_bf16(qkv, "qkv");
  const bool kv8 = kscale.has_value();
  if (kv8) {
    TORCH_CHECK(kpool.scalar_type() == torch::kUInt8 &&
                    vpool.scalar_type() == torch::kUInt8,
                "fp8 KV cache must be uint8 pools");
    check_f32(*kscale, "kscale");
    check_f32(*vscale, "vscale");
  } else {
    check_bf16(kpool, "kpool");
    check_bf16(vpool, "vpool");
  }
  check_f32(cos_t, "cos");
  check_f32(sin_t, "sin");
  const int n_layers_pool = (int)kpool.size(1);
  const int n_kv = (int)kpool.size(2);
  const int max_seq = (int)kpool.size(3);
  const int hs = (int)kpool.size(4);
  const int rope_ne = cos_t.dim() > 1 ? (int)cos_t.size(1) : 0;
  TORCH_CHECK(qkv.dim() == 2, "qkv must be [T, qkv_dim]");
  const int T = (int)qkv.size(0);
  const int qpk = (int)qkv.size(1) / (n_kv * hs) - 2;
  TORCH_CHECK((int64_t)pos0 + T <= max_seq, "prefill overflows the pool");
  launch_rope_prefill_append(
      qkv.data_ptr(), kpool.data_ptr(), vpool.data_ptr(),
      kv8 ? kscale->data_ptr<float>() : nullptr,
      kv8 ? vscale->data_ptr<float>() : nullptr,
      rope_ne ? cos_t.data_ptr<float>() : nullptr,
      rope_ne ? sin_t.data_ptr<float>() : nullptr, (int)pos0, (int)slot,
      (int)layer, n_layers_pool, n_kv, max_seq, hs, rope_ne, qpk, T,
      cur_stream());
}

void prefill_attn(torch::Tensor out, torch::Tensor qkv, torch::Tensor kpool,
                  torch::Tensor vpool, int64_t pos0, int64_t slot,
                  int64_t layer, double scale,
                  c10::optional<torch::Tensor> kscale,
                  c10::optional<torch::Tensor> vscale) {
  check_bf16(out, "out");
  check_bf16(qkv, "qkv");
  const bool kv8 = kscale.has_value();
  if (kv8) {
    TORCH_CHECK(kpool.scalar_type() == torch::kUInt8 &&
                    vpool.scalar_type() == torch::kUInt8,
                "fp8 KV cache must be uint8 pools");
    check_f32(*kscale, "kscale");
    check_f32(*vscale, "vscale");
  } else {
    check_bf16(kpool, "kpool");
    check_bf16(vpool, "vpool");
  }
  const int n_layers_pool = (int)kpool.size(1);
  const int n_kv = (int)kpool.size(2);
  const int max_seq = (int)kpool.size(3);
  const int hs = (int)kpool.size(4);
  const int T = (int)qkv.size(0);
  const int qpk = (int)qkv.size(1) / (n_kv * hs) - 2;
  TORCH_CHECK(out.numel() == (int64_t)T * n_kv * qpk * hs, "out size");
  int rc = launch_prefill_attn(
      out.data_ptr(), qkv.data_ptr(), kpool.data_ptr(), vpool.data_ptr(),
      kv8 ? kscale->data_ptr<float>() : nullptr,
      kv8 ? vscale->data_ptr<float>() : nullptr,
      (int)pos0, (int)slot, (int)layer, n_layers_pool, n_kv, max_seq, hs,
      qpk, T, (float)scale, cur_stream());
  TORCH_CHECK(rc == 0, "prefill_attn: unsupported head_size ", hs);
}

void add(torch::Tensor out, torch::Tensor a, torch::Tensor b) {
  check_bf16(out, "out");
  check_bf16(a, "a");
  check_bf16(b, "b");
  const int n = (int)a.numel();
  TORCH_CHECK(n % 8 == 0, "n must be a multiple of 8");
  launch_add(out.data_ptr(), a.data_ptr(), b.data_ptr(), n, cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "mdi_llm_amd hand-written CDNA4 (gfx950) decode kernels";
  m.def("rmsnorm", &rmsnorm, "RMSNorm (decode, bf16)");
  m.def("layernorm", &layernorm, "LayerNorm (decode, bf16)");
  m.def("gemv", &gemv, "decode GEMV out = W@norm?(x) (+bias)(+res)(act)",
        py::arg("out"), py::arg("W"), py::arg("x"), py::arg("bias"),
        py::arg("res"), py::arg("epilogue"),
        py::arg("norm_w") = c10::nullopt, py::arg("norm_b") = c10::nullopt,
        py::arg("norm_kind") = 0, py::arg("eps") = 1e-5,
        py::arg("rows") = 0, py::arg("eidx") = c10::nullopt,
        py::arg("estride") = 0);
  m.def("gemv_fp8", &gemv_fp8,
        "decode GEMV with fp8(e4m3) weights + per-row scales",
        py::arg("out"), py::arg("W"), py::arg("wscale"), py::arg("x"),
        py::arg("bias"), py::arg("res"), py::arg("epilogue"),
        py::arg("norm_w") = c10::nullopt, py::arg("norm_b") = c10::nullopt,
        py::arg("norm_kind") = 0, py::arg("eps") = 1e-5,
        py::arg("rows") = 0);
  m.def("gemv_swiglu_fp8", &gemv_swiglu_fp8,
        "fused SwiGLU pair GEMV with fp8 weights",
        py::arg("out"), py::arg("Wg"), py::arg("gscale"), py::arg("Wu"),
        py::arg("uscale"), py::arg("x"), py::arg("gelu_gate"),
        py::arg("norm_w") = c10::nullopt, py::arg("norm_b") = c10::nullopt,
        py::arg("norm_kind") = 0, py::arg("eps") = 1e-5);
  m.def("gemv_swiglu", &gemv_swiglu, "fused SwiGLU pair GEMV (+pre-norm)",
        py::arg("out"), py::arg("Wg"), py::arg("Wu"), py::arg("x"),
        py::arg("gelu_gate"), py::arg("norm_w") = c10::nullopt,
        py::arg("norm_b") = c10::nullopt, py::arg("norm_kind") = 0,
        py::arg("eps") = 1e-5, py::arg("eidx") = c10::nullopt,
        py::arg("estride") = 0, py::arg("escale") = c10::nullopt);
  m.def("swiglu_mul", &swiglu_mul,
        "batched act(gate)*up elementwise (out may alias up)",
        py::arg("out"), py::arg("g"), py::arg("u"), py::arg("gelu_gate"));
  m.def("moe_gate_topk", &moe_gate_topk,
        "MoE router: top-k experts + softmax weights over the k",
        py::arg("eidx"), py::arg("escale"), py::arg("logits"), py::arg("k"));
  m.def("embed", &embed, "embedding row gather");
  m.def("rope_kv_append", &rope_kv_append,
        "RoPE on interleaved qkv + KV cache append");
  m.def("attn_decode", &attn_decode,
        "GQA flash-decode attention (split-S, fused rope+append)",
        py::arg("out"), py::arg("part_o"), py::arg("part_ml"),
        py::arg("qkv"), py::arg("kpool"), py::arg("vpool"), py::arg("cos"),
        py::arg("sin"), py::arg("pos"), py::arg("slot"), py::arg("layer"),
        py::arg("n_chunks"), py::arg("scale"), py::arg("n_batch") = 0,
        py::arg("kscale") = c10::nullopt, py::arg("vscale") = c10::nullopt,
        py::arg("force_split") = 0);
  m.def("add", &add, "bf16 residual add");
  m.def("rope_prefill_append", &rope_prefill_append,
        "prefill: rope q/k for T positions + append k/v to the pool",
        py::arg("qkv"), py::arg("kpool"), py::arg("vpool"), py::arg("cos"),
        py::arg("sin"), py::arg("pos0"), py::arg("slot"), py::arg("layer"),
        py::arg("kscale") = c10::nullopt, py::arg("vscale") = c10::nullopt);
  m.def("prefill_attn", &prefill_attn,
        "causal GQA prefill flash attention (MFMA, online softmax)",
        py::arg("out"), py::arg("qkv"), py::arg("kpool"), py::arg("vpool"),
        py::arg("pos0"), py::arg("slot"), py::arg("layer"),
        py::arg("scale"), py::arg("kscale") = c10::nullopt,
        py::arg("vscale") = c10::nullopt);
  m.def("attn_proj", &attn_proj,
        "fused GQA flash-decode attention + output projection (one "
        "launch; in-launch granule hand-off overlaps the proj weight "
        "stream with attention)",
        py::arg("out"), py::arg("qkv"), py::arg("kpool"), py::arg("vpool"),
        py::arg("cos"), py::arg("sin"), py::arg("pos"), py::arg("slot"),
        py::arg("layer"), py::arg("scale"), py::arg("W"),
        py::arg("bias"), py::arg("res"), py::arg("gran"));
  m.def("sample", &sample,
        "fused temperature/top-k/gumbel token sampling (128k vocab ~15us)",
        py::arg("out_token"), py::arg("logits"), py::arg("scratch"),
        py::arg("temperature"), py::arg("top_k"), py::arg("noise"),
        py::arg("seed"), py::arg("pos") = c10::nullopt,
        py::arg("slot") = c10::nullopt, py::arg("n_batch") = 0,
        py::arg("top_p") = 1.0, py::arg("token_table") = c10::nullopt,
        py::arg("pos_table") = c10::nullopt,
        py::arg("adv_slot") = c10::nullopt, py::arg("adv_pos") = 0,
        py::arg("pos_bias") = 0);
  m.def("mtile_gemm", &mtile_gemm,
        "grouped-decode M-tile MFMA GEMM (bulk LDS-staged X, nt W stream)",
        py::arg("Y"), py::arg("W"), py::arg("X"), py::arg("bias"),
        py::arg("res"));
  m.def("route_env", &route_env,
        "device-side envelope routing (slot from header, dummy on stop)",
        py::arg("hdr"), py::arg("slot_out"), py::arg("pos_table"),
        py::arg("dummy_slot"));
  m.def("stage_slot", &stage_slot,
        "one-launch step staging/bookkeeping on device scalars",
        py::arg("slot"), py::arg("pos_out") = c10::nullopt,
        py::arg("token_out") = c10::nullopt,
        py::arg("pos_table") = c10::nullopt,
        py::arg("token_table") = c10::nullopt,
        py::arg("pos_table_mut") = c10::nullopt, py::arg("adv_pos") = 0);
}
