// Launcher declarations shared between the HIP kernel TU (hipcc) and the
// torch bindings TU (host C++). hipStream_t is an opaque pointer on both
// sides; the bindings include HIP headers for the real typedef.
#pragma once

#include <hip/hip_runtime.h>

// n_batch rows of [n] normalized independently (n_batch<=0 -> 1)
void launch_rmsnorm(void* out, const void* x, const void* w, int n, float eps,
                    int n_batch, hipStream_t stream);

void launch_layernorm(void* out, const void* x, const void* w, const void* b,
                      int n, float eps, hipStream_t stream);

// epilogue: 0 none, 1 +res, 2 gelu(tanh), 3 silu
// norm_kind: 0 none, 1 fused RMSNorm on x, 2 fused LayerNorm on x
// rows: output rows per wave (1/2/4); 0 = auto by M
// eidx/estride (optional): W += eidx[0]*estride at replay time — selects
// an expert slab from a stacked [n_expert, M, K] weight (MoE routing
// stays graph-capturable); norm_kind 2 does not support indirection
void launch_gemv(void* out, const void* W, const void* x, const void* bias,
                 const void* res, const void* norm_w, const void* norm_b,
                 float eps, int M, int K, int epilogue, int norm_kind,
                 int rows, const int* eidx, long long estride,
                 hipStream_t stream);

// fp8(e4m3)-weight variants: W is bytes, wscale fp32 per output row
void launch_gemv_fp8(void* out, const void* W, const float* wscale,
                     const void* x, const void* bias, const void* res,
                     const void* norm_w, const void* norm_b, float eps,
                     int M, int K, int epilogue, int norm_kind, int rows,
                     hipStream_t stream);

void launch_gemv_swiglu_fp8(void* out, const void* Wg, const float* gscale,
                            const void* Wu, const float* uscale,
                            const void* x, const void* norm_w,
                            const void* norm_b, float eps, int M, int K,
                            int gelu_gate, int norm_kind,
                            hipStream_t stream);

void launch_gemv_swiglu(void* out, const void* Wg, const void* Wu,
                        const void* x, const void* norm_w, const void* norm_b,
                        float eps, int M, int K, int gelu_gate, int norm_kind,
                        const int* eidx, long long estride,
                        const float* escale, hipStream_t stream);

// batched elementwise act(g)*u (out may alias u); n multiple of 8
void launch_swiglu_mul(void* out, const void* g, const void* u,
                       long long n, int gelu_gate, hipStream_t stream);

// MoE router: eidx/escale[k] = top-k experts of gate logits + softmax
// weights over the selected k (<= 8 of <= 64 experts)
void launch_moe_gate_topk(int* eidx, float* escale, const void* logits,
                          int n_e, int k, hipStream_t stream);

void launch_embed(void* out, const void* wte, const int* token, int n_embd,
                  float scale, hipStream_t stream);

void launch_rope_kv_append(void* qkv, void* kpool, void* vpool,
                           const float* cos_t, const float* sin_t,
                           const int* pos, const int* slot, int layer,
                           int n_layers_pool, int n_kv_heads, int max_seq,
                           int head_size, int rope_n_elem, int qpk,
                           hipStream_t stream);

// qkv is RAW (rope applied inside); the kernel also appends the current
// token's roped k and raw v into the pools at pos.
// n_batch == 0: single-token mode (pos/slot device scalars);
// n_batch > 0: batched mode (pos/slot arrays [n_batch], leading batch dim
// on qkv/part_o/part_ml/out).
int launch_attn_decode(void* out, float* part_o, float* part_ml,
                       const void* qkv, void* kpool, void* vpool,
                       float* kscale, float* vscale,
                       const float* cos_t, const float* sin_t, int rope_ne,
                       const int* pos, const int* slot, int layer,
                       int n_layers_pool, int n_kv_heads, int max_seq,
                       int head_size, int qpk, int n_chunks, float scale,
                       int n_batch, int force_split, hipStream_t stream);

void launch_add(void* out, const void* a, const void* b, int n,
                hipStream_t stream);

// fused temperature/top-k/top-p/gumbel sampling; scratch: >=520 u32 PER
// SAMPLE, zeroed initially (self-cleaning); n_batch draws from
// [n_batch, V] logits.  The gumbel stream is keyed by (seed, slot, pos) —
// fused attention + output-projection (one launch; granule hand-off
// inside).  Returns -1 when the geometry/max_seq has no instantiation.
int launch_attn_proj(void* out, const void* qkv, void* kpool, void* vpool,
                     const float* cos_t, const float* sin_t, int rope_ne,
                     const int* pos, const int* slot, int layer,
                     int n_layers_pool, int n_kv_heads, int max_seq,
                     int head_size, int qpk, float scale, const void* W,
                     const void* bias, const void* res, void* gran, int M,
                     hipStream_t stream);

// reproducible and schedule-independent.  When token_table/pos_table and
// adv_slot are given (single-sample mode), the unpack step also writes
// token_table[slot] and advances pos_table[slot] by adv_pos in the same
// launch (graph bookkeeping without extra kernels).
void launch_sample(void* out_token, const void* logits, int V, void* scratch,
                   float temperature, int top_k, float top_p, int noise_on,
                   unsigned seed, const int* pos, const int* slot,
                   int n_batch, int* token_table, int* pos_table,
                   const int* adv_slot, int adv_pos, int pos_bias,
                   hipStream_t stream);

// grouped M-tile MFMA GEMM: Y[B,M] = X[B,K] @ W[M,K]^T (+bias, +res);
// returns -1 when (B, K) has no instantiation
int launch_mtile_gemm(void* Y, const void* W, const void* X,
                      const void* bias, const void* res, int Bsz, int M,
                      int K, hipStream_t stream);

// envelope routing for the pipelined secondary serve: slot_out <-
// hdr[0] (data) or dummy_slot (stop/flush, with pos reset)
void launch_route_env(const int* hdr, int* slot_out, int* pos_table,
                      int dummy_slot, hipStream_t stream);

// one-launch step staging: pos_out = pos_table[slot], token_out =
// token_table[slot], optional pos_table_mut[slot] += 1; any output may be
// null
void launch_stage_slot(int* pos_out, int* token_out, const int* pos_table,
                       const int* token_table, int* pos_table_mut,
                       const int* slot, int adv_pos, hipStream_t stream);

// prefill: rope+append all T positions (grid covers T)
void launch_rope_prefill_append(void* qkv, void* kpool, void* vpool,
                                float* kscale, float* vscale,
                                const float* cos_t, const float* sin_t,
                                int pos0, int slot, int layer,
                                int n_layers_pool, int n_kv_heads,
                                int max_seq, int head_size, int rope_ne,
                                int qpk, int T, hipStream_t stream);

// prefill flash attention (causal, GQA); out [T, n_head*head_size]
int launch_prefill_attn(void* out, const void* qkv, const void* kpool,
                        const void* vpool, const float* kscale,
                        const float* vscale, int pos0, int slot, int layer,
                        int n_layers_pool, int n_kv_heads, int max_seq,
                        int head_size, int qpk, int T, float scale,
                        hipStream_t stream);
