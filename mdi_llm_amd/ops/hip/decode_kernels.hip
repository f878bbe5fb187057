// decode_kernels.hip — hand-written CDNA4 (gfx950 / MI355X) kernels for the
// per-token decode path of mdi_llm_amd.
//
// These implement, natively for MI355X, the ops the reference runs through
// PyTorch/cuBLAS (survey §2.4; /root/reference/src/sub/model.py): RMSNorm /
// LayerNorm, the decode GEMVs (fused-QKV, attention proj, SwiGLU MLP, GELU
// MLP, lm-head), RoPE + KV-cache append, and GQA flash-decode attention
// (MFMA 16x16x32 bf16 QK^T with online softmax, split-S across workgroups).
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * wave = 64 lanes; all block sizes are multiples of 64.
//  * bf16 global loads vectorized as int4 (16 B = 8 bf16 per lane).
//  * decode GEMV streams weights once; x is staged in LDS per block.
//  * attention uses the "swapped QK^T" MFMA so each lane owns whole
//    key-scores for one query head (softmax needs only small shfl groups).
//  * all sequence positions / sample slots arrive via device memory
//    (int32 tensors) so the whole decode step is hipGraph-replayable.

#include <cstdlib>
#include <type_traits>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include "decode_kernels.h"

#define DEVINL __device__ __forceinline__

using bf16 = __hip_bfloat16;

DEVINL float b2f(bf16 v) { return __bfloat162float(v); }
DEVINL bf16 f2b(float v) { return __float2bfloat16(v); }

// 8 bf16 loaded as one int4 (16 B)
struct bf16x8 {
  bf16 v[8];
};

DEVINL bf16x8 load8(const bf16* p) {
  bf16x8 r;
  *reinterpret_cast<int4*>(r.v) = *reinterpret_cast<const int4*>(p);
  return r;
}

// nontemporal variant for streamed-once weight reads (no L2 retention);
// the builtin needs a clang ext_vector type, not HIP's int4 struct
using i32x4_t = __attribute__((ext_vector_type(4))) int;

DEVINL bf16x8 load8_nt(const bf16* p) {
  bf16x8 r;
  *reinterpret_cast<i32x4_t*>(r.v) =
      __builtin_nontemporal_load(reinterpret_cast<const i32x4_t*>(p));
  return r;
}

DEVINL i32x4_t load16_nt_u8(const unsigned char* p) {
  return __builtin_nontemporal_load(reinterpret_cast<const i32x4_t*>(p));
}

using f32x4 = __attribute__((__vector_size__(16))) float;
using bf16x8_t = __attribute__((ext_vector_type(8))) __bf16;

DEVINL float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// ---- fp8 (OCP e4m3) KV cache -------------------------------------------
// Rows are stored per-row-scaled (one fp32 scale per cached (slot, layer,
// kv-head, position) row, absmax/448 — same scheme as the fp8 weights).
// gfx950 cvt builtins convert packed pairs; KV8 kernel variants load 8
// bytes instead of 16 per fragment and dequantize in registers.
using u8kv = unsigned char;
typedef float v2fkv __attribute__((ext_vector_type(2)));

DEVINL u8kv f32_to_fp8(float f) {
  const int packed = __builtin_amdgcn_cvt_pk_fp8_f32(f, 0.f, 0, false);
  return (u8kv)(packed & 0xFF);
}

DEVINL float fp8_to_f32(u8kv v) {
  return __builtin_amdgcn_cvt_f32_fp8((int)v, 0);
}

DEVINL bf16x8_t fp8x8_to_bf16(const u8kv* p, float scl) {
  const int lo = *reinterpret_cast<const int*>(p);
  const int hi = *reinterpret_cast<const int*>(p + 4);
  const v2fkv a = __builtin_amdgcn_cvt_pk_f32_fp8(lo, false);
  const v2fkv b = __builtin_amdgcn_cvt_pk_f32_fp8(lo, true);
  const v2fkv c = __builtin_amdgcn_cvt_pk_f32_fp8(hi, false);
  const v2fkv d = __builtin_amdgcn_cvt_pk_f32_fp8(hi, true);
  bf16x8_t r;
  r[0] = (__bf16)(a.x * scl);
  r[1] = (__bf16)(a.y * scl);
  r[2] = (__bf16)(b.x * scl);
  r[3] = (__bf16)(b.y * scl);
  r[4] = (__bf16)(c.x * scl);
  r[5] = (__bf16)(c.y * scl);
  r[6] = (__bf16)(d.x * scl);
  r[7] = (__bf16)(d.y * scl);
  return r;
}

// PV accumulate straight from the fp8 pairs (skips the bf16 round trip
// the MFMA-operand path needs)
DEVINL void fp8x8_fma(float* o, float pw, const u8kv* p, float scl) {
  const int lo = *reinterpret_cast<const int*>(p);
  const int hi = *reinterpret_cast<const int*>(p + 4);
  const v2fkv a = __builtin_amdgcn_cvt_pk_f32_fp8(lo, false);
  const v2fkv b = __builtin_amdgcn_cvt_pk_f32_fp8(lo, true);
  const v2fkv c = __builtin_amdgcn_cvt_pk_f32_fp8(hi, false);
  const v2fkv d = __builtin_amdgcn_cvt_pk_f32_fp8(hi, true);
  const float ps = pw * scl;
  o[0] += ps * a.x;
  o[1] += ps * a.y;
  o[2] += ps * b.x;
  o[3] += ps * b.y;
  o[4] += ps * c.x;
  o[5] += ps * c.y;
  o[6] += ps * d.x;
  o[7] += ps * d.y;
}

// atomicMax for a non-negative float held in LDS as its uint bits
DEVINL void lds_fmax_u(unsigned* addr, float v) {
  atomicMax(addr, __float_as_uint(v));
}

DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// ---------------------------------------------------------------------------
// RMSNorm (decode, one vector): out = x * rsqrt(mean(x^2)+eps) * w
// fp32 accumulation, matching the torch reference (model.py RMSNorm).
// One block (256 threads) per vector.
// ---------------------------------------------------------------------------
__global__ void rmsnorm_kernel(bf16* __restrict__ out,
                               const bf16* __restrict__ x,
                               const bf16* __restrict__ w, int n, float eps) {
  __shared__ float red[4];
  // batched: row blockIdx.y of [B, n]
  out += (size_t)blockIdx.y * n;
  x += (size_t)blockIdx.y * n;
  const int tid = threadIdx.x;
  float acc = 0.f;
  // vectorized: 8 bf16 per step
  for (int i = tid * 8; i < n; i += blockDim.x * 8) {
    bf16x8 v = load8(x + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = b2f(v.v[j]);
      acc += f * f;
    }
  }
  acc = wave_reduce_sum(acc);
  if ((tid & 63) == 0) red[tid >> 6] = acc;
  __syncthreads();
  float total = red[0] + red[1] + red[2] + red[3];
  float scale = rsqrtf(total / n + eps);
  for (int i = tid * 8; i < n; i += blockDim.x * 8) {
    bf16x8 v = load8(x + i);
    bf16x8 g = load8(w + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.v[j] = f2b(b2f(v.v[j]) * scale * b2f(g.v[j]));
    *reinterpret_cast<int4*>(out + i) = *reinterpret_cast<const int4*>(o.v);
  }
}

// ---------------------------------------------------------------------------
// LayerNorm (decode): out = (x-mean)/sqrt(var+eps) * w + b
// ---------------------------------------------------------------------------
__global__ void layernorm_kernel(bf16* __restrict__ out,
                                 const bf16* __restrict__ x,
                                 const bf16* __restrict__ w,
                                 const bf16* __restrict__ b, int n,
                                 float eps) {
  __shared__ float red[8];
  const int tid = threadIdx.x;
  float s = 0.f, s2 = 0.f;
  for (int i = tid * 8; i < n; i += blockDim.x * 8) {
    bf16x8 v = load8(x + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = b2f(v.v[j]);
      s += f;
      s2 += f * f;
    }
  }
  s = wave_reduce_sum(s);
  s2 = wave_reduce_sum(s2);
  if ((tid & 63) == 0) {
    red[(tid >> 6) * 2] = s;
    red[(tid >> 6) * 2 + 1] = s2;
  }
  __syncthreads();
  float mean = (red[0] + red[2] + red[4] + red[6]) / n;
  float var = (red[1] + red[3] + red[5] + red[7]) / n - mean * mean;
  float inv = rsqrtf(var + eps);
  for (int i = tid * 8; i < n; i += blockDim.x * 8) {
    bf16x8 v = load8(x + i);
    bf16x8 g = load8(w + i);
    bf16x8 o;
    if (b != nullptr) {
      bf16x8 bb = load8(b + i);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.v[j] = f2b((b2f(v.v[j]) - mean) * inv * b2f(g.v[j]) + b2f(bb.v[j]));
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.v[j] = f2b((b2f(v.v[j]) - mean) * inv * b2f(g.v[j]));
    }
    *reinterpret_cast<int4*>(out + i) = *reinterpret_cast<const int4*>(o.v);
  }
}

// ---------------------------------------------------------------------------
// Decode GEMV: out[M] = W[M,K] @ norm?(x)[K] (+ bias) (+ residual) (+ act)
//
// W row-major bf16, streamed once from HBM (the decode bound); x staged in
// LDS (K*2 bytes, up to ~28 KiB for the 14336-wide MLP). Block = 256
// threads = 4 waves; each wave owns TWO adjacent output rows per grid-stride
// step (doubles load ILP); lane reads 16 B per row per iteration.
//
// NORM fuses the pre-norm into the staging pass (saves a separate tiny
// kernel + a global round-trip per call): after staging raw x, the block
// reduces mean/meansq and rewrites LDS with the normalized+weighted value.
// NORM: 0 none, 1 RMSNorm, 2 LayerNorm.
// EPI:  0 none, 1 +residual, 2 gelu(tanh), 3 silu.
// ---------------------------------------------------------------------------
DEVINL float gelu_tanh(float v) {
  float c = 0.7978845608028654f * (v + 0.044715f * v * v * v);
  return 0.5f * v * (1.f + tanhf(c));
}

// Stage x into LDS, optionally fusing the pre-norm.  Returns a uniform
// post-scale to apply to each accumulated dot product:
//  NORM==1 (RMSNorm): LDS holds x*g; returned scale = rsqrt(mean(x^2)+eps),
//    exact because the RMS scale is uniform over j (applied post-dot).
//  NORM==2 (LayerNorm): LDS holds the fully normalized value (two-pass,
//    mean subtraction is not a uniform post-scale); returns 1.
template <int NORM>
DEVINL float stage_x(bf16* xs, const bf16* __restrict__ x,
                     const bf16* __restrict__ nw, const bf16* __restrict__ nb,
                     int K, float eps, float* red) {
  const int tid = threadIdx.x;
  if (NORM == 0) {
    for (int i = tid * 8; i < K; i += blockDim.x * 8)
      *reinterpret_cast<int4*>(xs + i) = *reinterpret_cast<const int4*>(x + i);
    __syncthreads();
    return 1.f;
  }
  if (NORM == 1) {
    float s2 = 0.f;
    for (int i = tid * 8; i < K; i += blockDim.x * 8) {
      bf16x8 v = load8(x + i);
      bf16x8 g = load8(nw + i);
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = b2f(v.v[j]);
        s2 += f * f;
        o.v[j] = f2b(f * b2f(g.v[j]));
      }
      *reinterpret_cast<int4*>(xs + i) = *reinterpret_cast<const int4*>(o.v);
    }
    s2 = wave_reduce_sum(s2);
    if ((tid & 63) == 0) red[tid >> 6] = s2;
    __syncthreads();
    float ms = (red[0] + red[1] + red[2] + red[3]) / K;
    return rsqrtf(ms + eps);
  }
  // NORM == 2: LayerNorm, two-pass
  float s = 0.f, s2 = 0.f;
  for (int i = tid * 8; i < K; i += blockDim.x * 8) {
    bf16x8 v = load8(x + i);
    *reinterpret_cast<int4*>(xs + i) = *reinterpret_cast<const int4*>(v.v);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = b2f(v.v[j]);
      s += f;
      s2 += f * f;
    }
  }
  s = wave_reduce_sum(s);
  s2 = wave_reduce_sum(s2);
  if ((tid & 63) == 0) {
    red[(tid >> 6) * 2] = s;
    red[(tid >> 6) * 2 + 1] = s2;
  }
  __syncthreads();
  const float mean = (red[0] + red[2] + red[4] + red[6]) / K;
  const float ms = (red[1] + red[3] + red[5] + red[7]) / K;
  const float inv = rsqrtf(ms - mean * mean + eps);
  for (int i = tid * 8; i < K; i += blockDim.x * 8) {
    bf16x8 v = load8(xs + i);
    bf16x8 g = load8(nw + i);
    bf16x8 o;
    if (nb != nullptr) {
      bf16x8 bb = load8(nb + i);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.v[j] = f2b((b2f(v.v[j]) - mean) * inv * b2f(g.v[j]) + b2f(bb.v[j]));
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.v[j] = f2b((b2f(v.v[j]) - mean) * inv * b2f(g.v[j]));
    }
    *reinterpret_cast<int4*>(xs + i) = *reinterpret_cast<const int4*>(o.v);
  }
  __syncthreads();
  return 1.f;
}

template <int EPI, int NORM, int ROWS>
__global__ void gemv_kernel(bf16* __restrict__ out,
                            const bf16* __restrict__ W,
                            const bf16* __restrict__ x,
                            const bf16* __restrict__ bias,
                            const bf16* __restrict__ res,
                            const bf16* __restrict__ nw,
                            const bf16* __restrict__ nb, float eps, int M,
                            int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __shared__ float red[8];
  bf16* xs = reinterpret_cast<bf16*>(smem);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int rows_per_grid = gridDim.x * (blockDim.x >> 6) * ROWS;
  const int row0 = (blockIdx.x * (blockDim.x >> 6) + wave) * ROWS;

  // Prefetch the first W chunk of the first row set BEFORE the staging
  // barrier: plain global loads stay in flight across s_barrier, so HBM
  // streams W while the block stages/normalizes x in LDS.
  bf16x8 wpre[ROWS];
  const bool pre_ok = lane * 8 < K;  // K is a multiple of 8
#pragma unroll
  for (int r = 0; r < ROWS; ++r)
    if (pre_ok)
      wpre[r] = load8_nt(W + (size_t)min(row0 + r, M - 1) * K + lane * 8);

  const float nscale = stage_x<NORM>(xs, x, nw, nb, K, eps, red);

  for (int row = row0; row < M; row += rows_per_grid) {
    const bf16* wrow[ROWS];
    float acc[ROWS];
#pragma unroll
    for (int r = 0; r < ROWS; ++r) {
      wrow[r] = W + (size_t)min(row + r, M - 1) * K;
      acc[r] = 0.f;
    }
    int istart = lane * 8;
    if (row == row0 && pre_ok) {
      // peeled first chunk: consume the pre-barrier prefetch
      bf16x8 xv = load8(xs + istart);
#pragma unroll
      for (int r = 0; r < ROWS; ++r) {
        bf16x8 wv;
        *reinterpret_cast<int4*>(wv.v) =
            *reinterpret_cast<const int4*>(wpre[r].v);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[r] += b2f(wv.v[j]) * b2f(xv.v[j]);
      }
      istart += 64 * 8;
    }
    for (int i = istart; i < K; i += 64 * 8) {
      bf16x8 xv = load8(xs + i);
#pragma unroll
      for (int r = 0; r < ROWS; ++r) {
        bf16x8 wv = load8_nt(wrow[r] + i);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[r] += b2f(wv.v[j]) * b2f(xv.v[j]);
      }
    }
#pragma unroll
    for (int r = 0; r < ROWS; ++r) acc[r] = wave_reduce_sum(acc[r]);
    if (lane == 0) {
#pragma unroll
      for (int r = 0; r < ROWS; ++r) {
        const int rw = row + r;
        if (rw >= M) break;
        float a = acc[r] * nscale;
        if (bias != nullptr) a += b2f(bias[rw]);
        if (EPI == 1 && res != nullptr) a += b2f(res[rw]);
        if (EPI == 2) a = gelu_tanh(a);
        if (EPI == 3) a = a / (1.f + expf(-a));
        out[rw] = f2b(a);
      }
    }
  }
}

// Direct-x variant (NORM 0/1 only): no LDS staging, no barrier — the W
// non-temporal stream starts at instruction one.  Each lane reads its x
// (and RMS weight) chunks straight from L1/L2 (x is KB-sized and hot),
// and because a wave's lanes partition the full K range, the RMSNorm
// mean-square reduces per-wave in registers (wave_reduce) with the exact
// post-dot scale trick — no cross-wave exchange at all.  Measured: the
// staged kernel spent ~1-4 us/launch filling LDS before W streaming
// ramped (proj/down ran at 4.4 TB/s in-graph vs 6.1 for the long swiglu
// kernel); this variant removes that ramp.  LayerNorm (NORM==2) keeps the
// staged two-pass form.
template <int EPI, int NORM, int ROWS, int UNR>
__global__ void gemv_direct_kernel(bf16* __restrict__ out,
                                   const bf16* __restrict__ W,
                                   const bf16* __restrict__ x,
                                   const bf16* __restrict__ bias,
                                   const bf16* __restrict__ res,
                                   const bf16* __restrict__ nw, float eps,
                                   int M, int K,
                                   const int* __restrict__ eidx,
                                   long long estride) {
  static_assert(NORM == 0 || NORM == 1, "direct gemv: no LayerNorm");
  if (eidx != nullptr) W += (size_t)eidx[0] * (size_t)estride;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int rows_per_grid = gridDim.x * (blockDim.x >> 6) * ROWS;
  const int row0 = (blockIdx.x * (blockDim.x >> 6) + wave) * ROWS;

  float nscale = 1.f;
  bool have_scale = false;
  for (int row = row0; row < M; row += rows_per_grid) {
    const bf16* wrow[ROWS];
    float acc[ROWS];
#pragma unroll
    for (int r = 0; r < ROWS; ++r) {
      wrow[r] = W + (size_t)min(row + r, M - 1) * K;
      acc[r] = 0.f;
    }
    float s2 = 0.f;
    // UNR independent W chunks in flight per lane.  Measured verdicts
    // (Llama-3-8B, rocprof per-kernel): NORM==0 short-K (proj 4096x4096)
    // 16.3 -> 8.7 us at UNR=4; NORM==1 (qkv, fused-RMS x*nw staging eats
    // the register headroom) and long-K (down, 28 iters) both REGRESS at
    // 4 — they stay at 2.  A register-staged fused-RMS variant (stage
    // x*nw once, stream W pure) also measured WORSE (22 us): its 1-row
    // waves pay the 16-load prologue without amortization.
    auto dot_chunk = [&](int i) {
      bf16x8 xv = load8(x + i);
      float xm[8];
      if (NORM == 1) {
        bf16x8 gv = load8(nw + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = b2f(xv.v[j]);
          s2 += f * f;
          xm[j] = f * b2f(gv.v[j]);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) xm[j] = b2f(xv.v[j]);
      }
#pragma unroll
      for (int r = 0; r < ROWS; ++r) {
        bf16x8 wv = load8_nt(wrow[r] + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[r] += b2f(wv.v[j]) * xm[j];
      }
    };
    if constexpr (UNR >= 4) {
#pragma unroll 4
      for (int i = lane * 8; i < K; i += 64 * 8) dot_chunk(i);
    } else {
#pragma unroll 2
      for (int i = lane * 8; i < K; i += 64 * 8) dot_chunk(i);
    }
#pragma unroll
    for (int r = 0; r < ROWS; ++r) acc[r] = wave_reduce_sum(acc[r]);
    if (NORM == 1 && !have_scale) {
      // lanes of this wave partitioned all of K: full mean-square
      nscale = rsqrtf(wave_reduce_sum(s2) / K + eps);
      have_scale = true;  // grid-stride re-entry reuses it
    }
    if (lane == 0) {
#pragma unroll
      for (int r = 0; r < ROWS; ++r) {
        const int rw = row + r;
        if (rw >= M) break;
        float a = acc[r] * nscale;
        if (bias != nullptr) a += b2f(bias[rw]);
        if (EPI == 1 && res != nullptr) a += b2f(res[rw]);
        if (EPI == 2) a = gelu_tanh(a);
        if (EPI == 3) a = a / (1.f + expf(-a));
        out[rw] = f2b(a);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Grouped-decode M-tile MFMA GEMM:  Y[B, M] = X[B, K] @ W[M, K]^T
// (+ bias / + residual), B in {16, 32, 64, 128}.
//
// The hipBLASLt skinny-M GEMMs run these shapes at ~68% of the weight-
// stream roofline (round-1 measurement); this is the priced recipe from
// the round-1 ROADMAP: X staged into LDS in BULK COALESCED chunks with
// register double-buffering (the naive per-fragment L2 X loads measured
// 1.4-1.6 TB/s — latency-bound), W streamed non-temporally straight into
// MFMA A-fragments.
//
// Geometry: one block per 16 output rows; its 4 waves split each K-chunk
// (intra-block split-K), partials reduced through LDS at the end.  MFMA
// 16x16x32 bf16 with the same fragment convention as the attention
// kernels: lane&15 = fragment row, (lane>>4)*8 = k offset; C[m, b] lives
// in lane (sub*4+r rows of column b = lane&15).
// X LDS tile is XOR-swizzled (row&7)<<4 bytes so ds_read_b128 at a
// power-of-two row stride stays bank-conflict-free.
// ---------------------------------------------------------------------------
template <int KC>
DEVINL int x_swz(int row, int k) {  // element index into a [B][KC] tile
  int byte = (row * KC + k) * 2;
  byte ^= (row & 7) << 4;
  return byte >> 1;
}

template <int B>
__global__ void mtile_gemm_kernel(bf16* __restrict__ Y,
                                  const bf16* __restrict__ W,
                                  const bf16* __restrict__ X,
                                  const bf16* __restrict__ bias,
                                  const bf16* __restrict__ res,
                                  int M, int K) {
  constexpr int KC = (B <= 32) ? 512 : (B <= 64 ? 256 : 128);
  constexpr int BT = B / 16;      // 16-wide b tiles
  constexpr int PRE = B * KC / 8 / 256;  // staged int4 loads per thread (8)
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* xbuf0 = reinterpret_cast<bf16*>(smem);
  bf16* xbuf1 = xbuf0 + B * KC;
  // c_red ([4 waves][16 rows][B] fp32, 16*B*16 bytes <= 32 KB) OVERLAYS
  // the X buffers: the tiles are dead once the last chunk is consumed,
  // so total LDS stays 2*B*KC*2 = 64 KB -> 2 blocks/CU
  float* c_red = reinterpret_cast<float*>(smem);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = (int)blockIdx.x * 16;
  const int arow = lane & 15;
  const int sub = lane >> 4;
  const int koff = sub * 8;
  const int nch = K / KC;  // launcher guarantees K % KC == 0

  f32x4 acc[BT];
#pragma unroll
  for (int t = 0; t < BT; ++t) acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

  // ---- bulk-coalesced stage of chunk 0 ----------------------------------
  // thread i covers X element block (b = i / (KC/8), k8 = i % (KC/8))
  {
#pragma unroll
    for (int u = 0; u < PRE; ++u) {
      const int i = tid + u * 256;
      const int b = i / (KC / 8);
      const int k8 = (i - b * (KC / 8)) * 8;
      bf16x8 xv = load8(X + (size_t)b * K + k8);
      // k8 is 16-byte aligned and the swizzle permutes whole 16-B units,
      // so the 8 elements land contiguously: one ds_write_b128
      *reinterpret_cast<int4*>(&xbuf0[x_swz<KC>(b, k8)]) =
          *reinterpret_cast<int4*>(xv.v);
    }
  }
  __syncthreads();

  const bf16* wrow = W + (size_t)(row0 + arow) * K;
  const int q0 = wave * (KC / 4);
  constexpr int SPC = KC / 4 / 32;  // mfma steps per chunk per wave

  // W fragment ring, TWO chunks deep: global loads stay in flight across
  // s_barrier, so the weight stream never drains at a chunk boundary
  // (the v1 form with unroll-2 inside the chunk measured 2-3.3 TB/s —
  // latency-bound exactly like the unpipelined GEMVs)
  bf16x8 wfr[2][SPC];
#pragma unroll
  for (int s = 0; s < SPC; ++s)
    wfr[0][s] = load8_nt(wrow + q0 + s * 32 + koff);
  if (1 < nch) {
#pragma unroll
    for (int s = 0; s < SPC; ++s)
      wfr[1][s] = load8_nt(wrow + KC + q0 + s * 32 + koff);
  }

  // chunk loop unrolled by TWO so the ring slot and LDS buffer selection
  // are compile-time (a runtime wfr[ci&1] dynamically indexes a register
  // array -> scratch spill; measured 1.7x slower than even the v1 form)
  auto chunk_body = [&](int ci, bf16x8 (&wc)[SPC], bf16* cur, bf16* nxt) {
    bf16x8 pre[PRE];
    if (ci + 1 < nch) {
      const int kb = (ci + 1) * KC;
#pragma unroll
      for (int u = 0; u < PRE; ++u) {
        const int i = tid + u * 256;
        const int b = i / (KC / 8);
        const int k8 = (i - b * (KC / 8)) * 8;
        pre[u] = load8(X + (size_t)b * K + kb + k8);
      }
    }
#pragma unroll
    for (int sI = 0; sI < SPC; ++sI) {
      const int kl = q0 + sI * 32;
      const bf16x8_t wf = *reinterpret_cast<const bf16x8_t*>(wc[sI].v);
#pragma unroll
      for (int t = 0; t < BT; ++t) {
        const bf16x8_t xf = *reinterpret_cast<const bf16x8_t*>(
            &cur[x_swz<KC>(t * 16 + arow, kl + koff)]);
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(wf, xf, acc[t],
                                                         0, 0, 0);
      }
    }
    if (ci + 2 < nch) {
      const size_t kb = (size_t)(ci + 2) * KC;
#pragma unroll
      for (int sI = 0; sI < SPC; ++sI)
        wc[sI] = load8_nt(wrow + kb + q0 + sI * 32 + koff);
    }
    if (ci + 1 < nch) {
#pragma unroll
      for (int u = 0; u < PRE; ++u) {
        const int i = tid + u * 256;
        const int b = i / (KC / 8);
        const int k8 = (i - b * (KC / 8)) * 8;
        *reinterpret_cast<int4*>(&nxt[x_swz<KC>(b, k8)]) =
            *reinterpret_cast<int4*>(pre[u].v);
      }
    }
    __syncthreads();
  };
  for (int ci = 0; ci < nch; ci += 2) {   // launcher guarantees nch even
    chunk_body(ci, wfr[0], xbuf0, xbuf1);
    chunk_body(ci + 1, wfr[1], xbuf1, xbuf0);
  }

  // ---- intra-block K reduce through LDS + epilogue ----------------------
#pragma unroll
  for (int t = 0; t < BT; ++t)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      c_red[(wave * 16 + sub * 4 + r) * B + t * 16 + arow] = acc[t][r];
  __syncthreads();
  // threads cover (row, b) pairs: 16*B values
  for (int i = tid; i < 16 * B; i += 256) {
    const int r = i / B;
    const int b = i - r * B;
    float v = c_red[(0 * 16 + r) * B + b] + c_red[(1 * 16 + r) * B + b] +
              c_red[(2 * 16 + r) * B + b] + c_red[(3 * 16 + r) * B + b];
    const int m = row0 + r;
    if (bias != nullptr) v += b2f(bias[m]);
    if (res != nullptr) v += b2f(res[(size_t)b * M + m]);
    Y[(size_t)b * M + m] = f2b(v);
  }
}

// Register-staged fused-RMSNorm GEMV (qkv shape): the direct-x NORM==1
// form interleaves x and norm-weight loads with the W stream every
// chunk, which keeps the W pipeline at ~3.5 TB/s regardless of unroll
// depth (measured).  Here the normalized x (bf16, one extra rounding —
// matches the torch path, which also materializes the norm in bf16) is
// staged into registers ONCE, and the row loop issues nothing but
// non-temporal W loads: MAXCH*16 B in flight per lane.
// LAUNCHED WITH A REDUCED GRID so each wave runs SEVERAL rows: the
// first version ran 1 row/wave and the 16-load staging prologue cost as
// much as the row itself (22 us vs 14.5 direct — measured, reverted);
// amortized over ~3 rows the prologue is ~15% overhead.
// K must be a multiple of 512 and <= MAXCH*512.
template <int MAXCH>
__global__ void gemv_direct_pre_kernel(bf16* __restrict__ out,
                                       const bf16* __restrict__ W,
                                       const bf16* __restrict__ x,
                                       const bf16* __restrict__ bias,
                                       const bf16* __restrict__ nw,
                                       float eps, int M, int K) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int rows_per_grid = gridDim.x * (blockDim.x >> 6);
  const int row0 = blockIdx.x * (blockDim.x >> 6) + wave;
  const int nch = K >> 9;

  bf16x8 xm[MAXCH];
  float s2 = 0.f;
#pragma unroll
  for (int cI = 0; cI < MAXCH; ++cI) {
    if (cI < nch) {
      const int i = cI * 512 + lane * 8;
      bf16x8 xv = load8(x + i);
      bf16x8 gv = load8(nw + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = b2f(xv.v[j]);
        s2 += f * f;
        xm[cI].v[j] = f2b(f * b2f(gv.v[j]));
      }
    }
  }
  const float nscale = rsqrtf(wave_reduce_sum(s2) / K + eps);

  for (int row = row0; row < M; row += rows_per_grid) {
    const bf16* wr = W + (size_t)row * K + lane * 8;
    float acc = 0.f;
#pragma unroll
    for (int cI = 0; cI < MAXCH; ++cI) {
      if (cI < nch) {
        bf16x8 wv = load8_nt(wr + cI * 512);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc += b2f(wv.v[j]) * b2f(xm[cI].v[j]);
      }
    }
    acc = wave_reduce_sum(acc) * nscale;
    if (lane == 0) {
      if (bias != nullptr) acc += b2f(bias[row]);
      out[row] = f2b(acc);
    }
  }
}

// MoE router: top-k of the gate logits (<= 64 experts) + softmax over
// the selected k (reference model.py:823-853 semantics: topk first, then
// softmax over the k logits).  One wave; outputs device-side so a
// captured graph replays the routing data-dependently.
__global__ void moe_gate_topk_kernel(int* __restrict__ eidx,
                                     float* __restrict__ escale,
                                     const bf16* __restrict__ logits,
                                     int n_e, int k) {
  const int lane = threadIdx.x;
  float cur = (lane < n_e) ? b2f(logits[lane]) : -1e30f;
  float sel[8];
  int seli[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    if (j >= k) continue;
    float m = cur;
    int mi = lane;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float om = __shfl_xor(m, off, 64);
      const int oi = __shfl_xor(mi, off, 64);
      if (om > m || (om == m && oi < mi)) {
        m = om;
        mi = oi;
      }
    }
    sel[j] = m;
    seli[j] = mi;
    if (lane == mi) cur = -1e30f;
  }
  if (lane == 0) {
    const float mx = sel[0];
    float ssum = 0.f;
    float e[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      e[j] = (j < k) ? __expf(sel[j] - mx) : 0.f;
      ssum += e[j];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (j < k) {
        eidx[j] = seli[j];
        escale[j] = e[j] / ssum;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// FP8 (OCP e4m3) weight variants: W stored as fp8 with one fp32 scale per
// output row (absmax/448 quantization).  Halves the decode weight stream;
// dequant via the gfx950 packed converts (v_cvt_pk_f32_fp8).
// ---------------------------------------------------------------------------
using f32x2 = __attribute__((__vector_size__(8))) float;

DEVINL float dot16_fp8(const unsigned* wq, const bf16* xs) {
  float acc = 0.f;
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(wq[q], false);
    const f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(wq[q], true);
    const int b = q * 4;
    acc += lo[0] * b2f(xs[b + 0]) + lo[1] * b2f(xs[b + 1]) +
           hi[0] * b2f(xs[b + 2]) + hi[1] * b2f(xs[b + 3]);
  }
  return acc;
}

template <int EPI, int NORM, int ROWS>
__global__ void gemv_fp8_kernel(bf16* __restrict__ out,
                                const unsigned char* __restrict__ W,
                                const float* __restrict__ wscale,  // [M]
                                const bf16* __restrict__ x,
                                const bf16* __restrict__ bias,
                                const bf16* __restrict__ res,
                                const bf16* __restrict__ nw,
                                const bf16* __restrict__ nb, float eps,
                                int M, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __shared__ float red[8];
  bf16* xs = reinterpret_cast<bf16*>(smem);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int rows_per_grid = gridDim.x * (blockDim.x >> 6) * ROWS;
  const int row0 = (blockIdx.x * (blockDim.x >> 6) + wave) * ROWS;

  // prefetch across the staging barrier (16 fp8 = one dwordx4 per lane)
  i32x4_t wpre[ROWS];
  const bool pre_ok = lane * 16 < K;
#pragma unroll
  for (int r = 0; r < ROWS; ++r)
    if (pre_ok)
      wpre[r] = load16_nt_u8(W + (size_t)min(row0 + r, M - 1) * K +
                             lane * 16);

  const float nscale = stage_x<NORM>(xs, x, nw, nb, K, eps, red);

  for (int row = row0; row < M; row += rows_per_grid) {
    const unsigned char* wrow[ROWS];
    float acc[ROWS];
#pragma unroll
    for (int r = 0; r < ROWS; ++r) {
      wrow[r] = W + (size_t)min(row + r, M - 1) * K;
      acc[r] = 0.f;
    }
    int istart = lane * 16;
    if (row == row0 && pre_ok) {
#pragma unroll
      for (int r = 0; r < ROWS; ++r)
        acc[r] += dot16_fp8(reinterpret_cast<const unsigned*>(&wpre[r]),
                            xs + istart);
      istart += 64 * 16;
    }
    for (int i = istart; i < K; i += 64 * 16) {
#pragma unroll
      for (int r = 0; r < ROWS; ++r) {
        const i32x4_t wq = load16_nt_u8(wrow[r] + i);
        acc[r] += dot16_fp8(reinterpret_cast<const unsigned*>(&wq), xs + i);
      }
    }
#pragma unroll
    for (int r = 0; r < ROWS; ++r) acc[r] = wave_reduce_sum(acc[r]);
    if (lane == 0) {
#pragma unroll
      for (int r = 0; r < ROWS; ++r) {
        const int rw = row + r;
        if (rw >= M) break;
        float a = acc[r] * nscale * wscale[rw];
        if (bias != nullptr) a += b2f(bias[rw]);
        if (EPI == 1 && res != nullptr) a += b2f(res[rw]);
        if (EPI == 2) a = gelu_tanh(a);
        if (EPI == 3) a = a / (1.f + expf(-a));
        out[rw] = f2b(a);
      }
    }
  }
}

template <int NORM>
__global__ void gemv_swiglu_fp8_kernel(
    bf16* __restrict__ out, const unsigned char* __restrict__ Wg,
    const float* __restrict__ gscale, const unsigned char* __restrict__ Wu,
    const float* __restrict__ uscale, const bf16* __restrict__ x,
    const bf16* __restrict__ nw, const bf16* __restrict__ nb, float eps,
    int M, int K, int gelu_gate) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __shared__ float red[8];
  bf16* xs = reinterpret_cast<bf16*>(smem);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int waves_per_grid = gridDim.x * (blockDim.x >> 6);
  const int row0 = blockIdx.x * (blockDim.x >> 6) + wave;

  const float nscale = stage_x<NORM>(xs, x, nw, nb, K, eps, red);

  for (int row = row0; row < M; row += waves_per_grid) {
    const unsigned char* grow = Wg + (size_t)row * K;
    const unsigned char* urow = Wu + (size_t)row * K;
    float ga = 0.f, ua = 0.f;
    for (int i = lane * 16; i < K; i += 64 * 16) {
      const i32x4_t gq = load16_nt_u8(grow + i);
      const i32x4_t uq = load16_nt_u8(urow + i);
      ga += dot16_fp8(reinterpret_cast<const unsigned*>(&gq), xs + i);
      ua += dot16_fp8(reinterpret_cast<const unsigned*>(&uq), xs + i);
    }
    ga = wave_reduce_sum(ga) * nscale;
    ua = wave_reduce_sum(ua) * nscale;
    if (lane == 0) {
      ga *= gscale[row];
      ua *= uscale[row];
      float act = gelu_gate ? gelu_tanh(ga) : ga / (1.f + expf(-ga));
      out[row] = f2b(act * ua);
    }
  }
}

// SwiGLU pair GEMV: out[i] = silu(Wg_i . xn) * (Wu_i . xn), optional fused
// pre-norm like gemv_kernel.  eidx/estride select an expert weight slab
// from stacked [n_expert, M, K] tensors AT LAUNCH-REPLAY TIME (the index
// lives in device memory, so MoE routing stays inside a captured graph);
// escale multiplies the output by the routing weight.
template <int NORM>
__global__ void gemv_swiglu_kernel(bf16* __restrict__ out,
                                   const bf16* __restrict__ Wg,
                                   const bf16* __restrict__ Wu,
                                   const bf16* __restrict__ x,
                                   const bf16* __restrict__ nw,
                                   const bf16* __restrict__ nb, float eps,
                                   int M, int K, int gelu_gate,
                                   const int* __restrict__ eidx,
                                   long long estride,
                                   const float* __restrict__ escale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __shared__ float red[8];
  bf16* xs = reinterpret_cast<bf16*>(smem);
  if (eidx != nullptr) {
    const size_t off = (size_t)eidx[0] * (size_t)estride;
    Wg += off;
    Wu += off;
  }
  const float oscale = (escale != nullptr) ? escale[0] : 1.f;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int waves_per_grid = gridDim.x * (blockDim.x >> 6);
  const int row0 = blockIdx.x * (blockDim.x >> 6) + wave;

  // W prefetch across the staging barrier (see gemv_kernel)
  const int rp0 = min(row0, M - 1);
  const bool pre_ok = lane * 8 < K;
  bf16x8 gpre, upre;
  if (pre_ok) {
    gpre = load8_nt(Wg + (size_t)rp0 * K + lane * 8);
    upre = load8_nt(Wu + (size_t)rp0 * K + lane * 8);
  }

  const float nscale = stage_x<NORM>(xs, x, nw, nb, K, eps, red);

  for (int row = row0; row < M; row += waves_per_grid) {
    const bf16* grow = Wg + (size_t)row * K;
    const bf16* urow = Wu + (size_t)row * K;
    float ga = 0.f, ua = 0.f;
    int istart = lane * 8;
    if (row == row0 && pre_ok) {
      bf16x8 xv = load8(xs + istart);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = b2f(xv.v[j]);
        ga += b2f(gpre.v[j]) * xf;
        ua += b2f(upre.v[j]) * xf;
      }
      istart += 64 * 8;
    }
    for (int i = istart; i < K; i += 64 * 8) {
      bf16x8 gv = load8_nt(grow + i);
      bf16x8 uv = load8_nt(urow + i);
      bf16x8 xv = load8(xs + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = b2f(xv.v[j]);
        ga += b2f(gv.v[j]) * xf;
        ua += b2f(uv.v[j]) * xf;
      }
    }
    ga = wave_reduce_sum(ga) * nscale;
    ua = wave_reduce_sum(ua) * nscale;
    if (lane == 0) {
      float act = gelu_gate ? gelu_tanh(ga) : ga / (1.f + expf(-ga));
      out[row] = f2b(act * ua * oscale);
    }
  }
}

// Batched SwiGLU glue for the grouped engine: out = act(g) * u over n
// elements (replaces a silu + mul torch pair = two launches per layer).
// out may alias u (pure elementwise).
__global__ void swiglu_mul_kernel(bf16* __restrict__ out,
                                  const bf16* __restrict__ g,
                                  const bf16* __restrict__ u,
                                  long long n, int gelu_gate) {
  const long long i0 = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long i = i0; i < n; i += stride) {
    bf16x8 gv = load8(g + i);
    bf16x8 uv = load8(u + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = b2f(gv.v[j]);
      const float act = gelu_gate ? gelu_tanh(gf)
                                  : gf / (1.f + __expf(-gf));
      o.v[j] = f2b(act * b2f(uv.v[j]));
    }
    *reinterpret_cast<int4*>(out + i) = *reinterpret_cast<const int4*>(o.v);
  }
}

// ---------------------------------------------------------------------------
// Embedding row gather: out[n_embd] = wte[token] (* scale)
// token id read from device memory (graph-replayable).
// ---------------------------------------------------------------------------
__global__ void embed_kernel(bf16* __restrict__ out,
                             const bf16* __restrict__ wte,
                             const int* __restrict__ token, int n_embd,
                             float scale) {
  const int t = token[0];
  const bf16* row = wte + (size_t)t * n_embd;
  for (int i = (blockIdx.x * blockDim.x + threadIdx.x) * 8; i < n_embd;
       i += gridDim.x * blockDim.x * 8) {
    bf16x8 v = load8(row + i);
    if (scale != 1.f) {
#pragma unroll
      for (int j = 0; j < 8; ++j) v.v[j] = f2b(b2f(v.v[j]) * scale);
    }
    *reinterpret_cast<int4*>(out + i) = *reinterpret_cast<const int4*>(v.v);
  }
}

// ---------------------------------------------------------------------------
// RoPE (rotate-half) on the fused interleaved QKV vector + KV-cache append.
//
// qkv layout (litGPT interleaved, model.py:686-718): per query group g:
//   [q_0..q_{qpk-1}, k, v] rows of head_size each.
// K/V pool layout: [slot, layer, kv_head, max_seq, head_size] bf16.
// position and slot are device int32 scalars.
// Grid: n_query_groups blocks, 256 threads; rope_n_elem may be < head_size
// (partial rotary, e.g. NeoX 25%).
// ---------------------------------------------------------------------------
__global__ void rope_kv_append_kernel(
    bf16* __restrict__ qkv, bf16* __restrict__ kpool,
    bf16* __restrict__ vpool, const float* __restrict__ cos_t,
    const float* __restrict__ sin_t, const int* __restrict__ pos_p,
    const int* __restrict__ slot_p, int layer, int n_layers_pool,
    int n_kv_heads, int max_seq, int head_size, int rope_n_elem, int qpk) {
  const int g = blockIdx.x;  // query group == kv head
  const int pos = pos_p[0];
  const int slot = slot_p[0];
  const int tid = threadIdx.x;
  const int half = rope_n_elem >> 1;

  bf16* base = qkv + (size_t)g * (qpk + 2) * head_size;

  // rope on q rows and the k row (qpk+1 rows)
  for (int idx = tid; idx < (qpk + 1) * half; idx += blockDim.x) {
    const int r = idx / half;
    const int d = idx % half;
    bf16* row = base + (size_t)r * head_size;
    float x1 = b2f(row[d]);
    float x2 = b2f(row[d + half]);
    float c1 = cos_t[(size_t)pos * rope_n_elem + d];
    float s1 = sin_t[(size_t)pos * rope_n_elem + d];
    float c2 = cos_t[(size_t)pos * rope_n_elem + d + half];
    float s2 = sin_t[(size_t)pos * rope_n_elem + d + half];
    row[d] = f2b(x1 * c1 - x2 * s1);
    row[d + half] = f2b(x2 * c2 + x1 * s2);
  }
  __syncthreads();

  // append k,v rows into the pool at (slot, layer, g, pos)
  const size_t cache_off =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
          (size_t)max_seq * head_size +
      (size_t)pos * head_size;
  const bf16* krow = base + (size_t)qpk * head_size;
  const bf16* vrow = base + (size_t)(qpk + 1) * head_size;
  for (int d = tid; d < head_size; d += blockDim.x) {
    kpool[cache_off + d] = krow[d];
    vpool[cache_off + d] = vrow[d];
  }
}

// ---------------------------------------------------------------------------
// GQA flash-decode attention. Two variants share the MFMA tile core:
//
//  * attn_decode_block_kernel — used when max_seq <= 4096 (chosen
//    statically at launch, so hipGraphs stay shape-stable): ONE block per
//    (batch, kv_head); its 4 waves split the keys tile-interleaved, each
//    wave's online-softmax state (m, l) lives in registers keyed to the
//    lane's query head and every cross-lane exchange is a shfl, so the
//    key loop has NO LDS traffic and NO barriers; the 4 per-wave partials
//    are combined through LDS by the whole block at the end. The entire
//    attention step is one launch with zero global scratch — at decode
//    context lengths both the old split-S kernel and its combine kernel
//    sat at the ~5 us in-stream launch floor, and a fence-based in-kernel
//    combine was measured SLOWER (agent-scope release = cross-XCD L2
//    writeback per wave); block-local combine avoids both.
//
//  * attn_decode_kernel (split-S) — for long contexts (max_seq > 4096):
//    one wave per (kv_head, chunk) over n_chunks chunks for full-chip
//    parallelism, fp32 partials to global, attn_combine_kernel reduces.
//
// MFMA 16x16x32 bf16 computes scores[key, qhead] (swapped operands so
// each lane owns 4 keys of ONE query head). qpk (query heads per kv
// head) <= 16.
// ---------------------------------------------------------------------------
#define ATTN_WAVES 4  // waves per block, each fully independent
// max_seq at/below which the one-launch block-local variant is used; the
// choice is static per engine (max_seq), so hipGraph shapes never change
#define ATTN_BLOCK_MAX_SEQ 4096

// XOR swizzle on the q LDS tile (rows at a power-of-two byte stride would
// otherwise put a ds_read_b128 lane group on one bank slot — guide §6 G4):
// byte ^= (row&7)<<4
template <int HS>
DEVINL int q_swz(int row, int d) {  // element index into a [16][HS] tile
  int byte = (row * HS + d) * 2;
  byte ^= (row & 7) << 4;
  return byte >> 1;
}

// rotate-half rope of one element d of a head row (raw pointer into qkv)
DEVINL float rope_elem(const bf16* row, int d, int ne,
                       const float* __restrict__ cos_t,
                       const float* __restrict__ sin_t, int pos) {
  float x = b2f(row[d]);
  if (d >= ne) return x;
  const int half = ne >> 1;
  const float c = cos_t[(size_t)pos * ne + d];
  const float s = sin_t[(size_t)pos * ne + d];
  if (d < half) return x * c - b2f(row[d + half]) * s;
  return x * c + b2f(row[d - half]) * s;
}

// Block-local variant: grid.x = (n_batch) * n_kv_heads, block = 256.
// BATCH semantics as in attn_decode_kernel below.
template <int QPK, int HS, int BATCH, int KV8>
__global__ void attn_decode_block_kernel(
    bf16* __restrict__ out,       // [B?, n_head * head_size]
    const bf16* __restrict__ qkv, // [B?, qkv_dim] interleaved, RAW
    void* __restrict__ kpool_v, void* __restrict__ vpool_v,
    float* __restrict__ kscale, float* __restrict__ vscale,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    int rope_ne, const int* __restrict__ pos_p,
    const int* __restrict__ slot_p, int layer, int n_layers_pool,
    int n_kv_heads, int max_seq, float scale, int n_batch) {
  using kvt = std::conditional_t<KV8 != 0, u8kv, bf16>;
  kvt* kpool = reinterpret_cast<kvt*>(kpool_v);
  kvt* vpool = reinterpret_cast<kvt*>(vpool_v);
  constexpr int head_size = HS;
  static_assert(HS % 32 == 0 && HS * QPK >= 64, "unsupported attn geometry");
  constexpr int ODIM = HS * QPK / 64;  // output dims per lane (PV map)

  // shared q tile + current k row (staged once per block) and the 4
  // per-wave partials for the final in-block combine
  __shared__ __attribute__((aligned(16))) bf16 q_lds[16 * HS];
  __shared__ __attribute__((aligned(16))) bf16 k_cur[HS];
  __shared__ float o_part[ATTN_WAVES][QPK][HS];
  __shared__ float ml_part[ATTN_WAVES][QPK][2];
  __shared__ unsigned kvmax[2];  // KV8: row absmax bits (k, v)

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int b = BATCH ? (int)blockIdx.x / n_kv_heads : 0;
  const int g = BATCH ? (int)blockIdx.x % n_kv_heads : (int)blockIdx.x;
  if (b >= n_batch) return;

  const int S = pos_p[b] + 1;
  const int pos = S - 1;
  const int slot = slot_p[b];
  const int n_head_all = n_kv_heads * QPK;
  if (BATCH) {
    const int qkv_dim = n_kv_heads * (QPK + 2) * head_size;
    qkv += (size_t)b * qkv_dim;
    out += (size_t)b * n_head_all * head_size;
  }

  const size_t cache_base =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
      (size_t)max_seq * head_size;
  const size_t scl_base =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
      (size_t)max_seq;
  const bf16* krow_cur = qkv + ((size_t)g * (QPK + 2) + QPK) * head_size;
  const bf16* vrow_cur = krow_cur + head_size;
  if (KV8 && threadIdx.x < 2) kvmax[threadIdx.x] = 0;

  // ---- block-cooperative staging: q rows (roped), current k row ----------
  // the swizzle permutes 16-byte units WITHIN a row, so the pad rows
  // (>= QPK) can be zero-filled with vector stores disjoint from the
  // roped real rows — no barrier needed between the two loops
  for (int i = threadIdx.x; i < (16 - QPK) * head_size / 8; i += 256)
    reinterpret_cast<bf16x8_t*>(q_lds + QPK * head_size)[i] = bf16x8_t{};
  for (int i = threadIdx.x; i < QPK * head_size; i += 256) {
    const int r = i / head_size;
    const int d = i % head_size;
    const bf16* qrow = qkv + ((size_t)g * (QPK + 2) + r) * head_size;
    q_lds[q_swz<HS>(r, d)] =
        f2b(rope_elem(qrow, d, rope_ne, cos_t, sin_t, pos));
  }
  for (int d = threadIdx.x; d < head_size; d += 256)
    k_cur[d] = f2b(rope_elem(krow_cur, d, rope_ne, cos_t, sin_t, pos));
  __syncthreads();  // the only barrier before the combine

  // append the current token's k,v (no other block touches (slot, g))
  if constexpr (KV8) {
    float lk = 0.f, lv = 0.f;
    for (int d = threadIdx.x; d < head_size; d += 256) {
      lk = fmaxf(lk, fabsf(b2f(k_cur[d])));
      lv = fmaxf(lv, fabsf(b2f(vrow_cur[d])));
    }
    lds_fmax_u(&kvmax[0], lk);
    lds_fmax_u(&kvmax[1], lv);
    __syncthreads();
    const float ks = fmaxf(__uint_as_float(kvmax[0]), 1e-12f) / 448.f;
    const float vs = fmaxf(__uint_as_float(kvmax[1]), 1e-12f) / 448.f;
    for (int d = threadIdx.x; d < head_size; d += 256) {
      kpool[cache_base + (size_t)pos * head_size + d] =
          f32_to_fp8(b2f(k_cur[d]) / ks);
      vpool[cache_base + (size_t)pos * head_size + d] =
          f32_to_fp8(b2f(vrow_cur[d]) / vs);
    }
    if (threadIdx.x == 0) {
      kscale[scl_base + pos] = ks;
      vscale[scl_base + pos] = vs;
    }
  } else {
    for (int d = threadIdx.x; d < head_size; d += 256) {
      kpool[cache_base + (size_t)pos * head_size + d] = k_cur[d];
      vpool[cache_base + (size_t)pos * head_size + d] = vrow_cur[d];
    }
  }

  // ---- per-lane roles ----------------------------------------------------
  // softmax map (A): qhead qa = lane & 15, key sub-row = lane >> 4
  // PV map (B): qb = lane % QPK, dim slice d0, ODIM dims per lane
  const int qa = lane & 15;
  const int sub = lane >> 4;
  const int qb = lane % QPK;
  const int d0 = (lane / QPK) * ODIM;

  // every lane redundantly tracks (m, l) of ITS query head qa — the 4
  // lanes sharing a qa stay in sync because they see identical reduced
  // values, so no LDS round-trips are needed in the loop
  float m_st = -1e30f, l_st = 0.f;
  float o_acc[ODIM];
#pragma unroll
  for (int i = 0; i < ODIM; ++i) o_acc[i] = 0.f;

  const int n_tiles = (S + 15) / 16;
  for (int t = wave; t < n_tiles; t += ATTN_WAVES) {
    const int key0 = t * 16;

    // ---- QK^T via MFMA: A = K tile (16 keys x 32 dims), B = q^T ----------
    f32x4 acc4 = {0.f, 0.f, 0.f, 0.f};
    const int arow = lane & 15;
    const int koff = (lane >> 4) * 8;
    const int akey = key0 + arow;
    const kvt* krow = kpool + cache_base + (size_t)akey * head_size;
    const bool row_valid = akey < S;
    const bool row_cur = akey == pos;
    float kscl = 1.f;
    if (KV8 && row_valid && !row_cur) kscl = kscale[scl_base + akey];
#pragma unroll
    for (int c = 0; c < HS / 32; ++c) {
      bf16x8_t af = {};
      if (row_cur)
        af = *reinterpret_cast<const bf16x8_t*>(&k_cur[c * 32 + koff]);
      else if (row_valid) {
        if constexpr (KV8)
          af = fp8x8_to_bf16(
              reinterpret_cast<const u8kv*>(krow) + c * 32 + koff, kscl);
        else
          af = *reinterpret_cast<const bf16x8_t*>(
              reinterpret_cast<const bf16*>(krow) + c * 32 + koff);
      }
      const bf16x8_t bq = *reinterpret_cast<const bf16x8_t*>(
          &q_lds[q_swz<HS>(arow, c * 32 + koff)]);
      acc4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bq, acc4, 0, 0, 0);
    }
    float sc[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = key0 + sub * 4 + r;
      sc[r] = (key < S) ? acc4[r] * scale : -1e30f;
    }

    // ---- online softmax, all state in registers, exchanges via shfl ------
    float tmax = fmaxf(fmaxf(sc[0], sc[1]), fmaxf(sc[2], sc[3]));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float m_new = fmaxf(m_st, tmax);
    const float alpha = __expf(m_st - m_new);
    m_st = m_new;
    float e[4];
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = key0 + sub * 4 + r;
      e[r] = (key < S) ? __expf(sc[r] - m_new) : 0.f;
      psum += e[r];
    }
    psum += __shfl_xor(psum, 16, 64);
    psum += __shfl_xor(psum, 32, 64);
    l_st = l_st * alpha + psum;

    // ---- P*V: o[qb][d0..] += p[s][qb] * V[s][d0..] -----------------------
    // p for (key s, head qb) lives in lane ((s>>2)<<4 | qb), register s&3
    const float alphaB = __shfl(alpha, qb, 64);
#pragma unroll
    for (int i = 0; i < ODIM; ++i) o_acc[i] *= alphaB;
    // fixed 16-iteration loop so it fully unrolls (e[s&3] must stay in
    // registers); keys past S have e == 0, their V row is clamped to pos
#pragma unroll
    for (int s = 0; s < 16; ++s) {
      const float p = __shfl(e[s & 3], ((s >> 2) << 4) | qb, 64);
      const int skey = min(key0 + s, pos);
      const bool vcur = skey == pos;
      const bf16* vrow_b = vrow_cur + d0;
      const kvt* vrow_p = vpool + cache_base + (size_t)skey * head_size + d0;
      float vscl = 1.f;
      if (KV8 && !vcur) vscl = vscale[scl_base + skey];
      if constexpr (ODIM >= 8) {
#pragma unroll
        for (int i = 0; i < ODIM; i += 8) {
          if (!vcur) {
            if constexpr (KV8) {
              fp8x8_fma(&o_acc[i], p,
                        reinterpret_cast<const u8kv*>(vrow_p) + i, vscl);
              continue;
            }
          }
          bf16x8_t vv;
          if (vcur) {
            vv = *reinterpret_cast<const bf16x8_t*>(vrow_b + i);
          } else {
            vv = *reinterpret_cast<const bf16x8_t*>(
                reinterpret_cast<const bf16*>(vrow_p) + i);
          }
#pragma unroll
          for (int j = 0; j < 8; ++j) o_acc[i + j] += p * (float)vv[j];
        }
      } else {
        // narrow ODIM (small head geometries): per-element loads
#pragma unroll
        for (int j = 0; j < ODIM; ++j) {
          float v;
          if (vcur)
            v = b2f(vrow_b[j]);
          else if constexpr (KV8)
            v = fp8_to_f32(reinterpret_cast<const u8kv*>(vrow_p)[j]) * vscl;
          else
            v = b2f(reinterpret_cast<const bf16*>(vrow_p)[j]);
          o_acc[j] += p * v;
        }
      }
    }
  }

  // ---- per-wave partials to LDS, then block combine ----------------------
  // (a wave that got no tiles contributes m=-inf, l=0, o=0 — weight 0)
#pragma unroll
  for (int i = 0; i < ODIM; ++i) o_part[wave][qb][d0 + i] = o_acc[i];
  if (lane < QPK) {
    ml_part[wave][lane][0] = m_st;  // lane < 16 => qa == lane
    ml_part[wave][lane][1] = l_st;
  }
  __syncthreads();

  for (int i = threadIdx.x; i < QPK * head_size; i += 256) {
    const int h = i / head_size;
    const int d = i % head_size;
    float M = -1e30f;
#pragma unroll
    for (int w = 0; w < ATTN_WAVES; ++w)
      M = fmaxf(M, ml_part[w][h][0]);
    float den = 0.f, num = 0.f;
#pragma unroll
    for (int w = 0; w < ATTN_WAVES; ++w) {
      const float wgt = __expf(ml_part[w][h][0] - M);
      den += wgt * ml_part[w][h][1];
      num += wgt * o_part[w][h][d];
    }
    out[(size_t)(g * QPK + h) * head_size + d] = f2b(num / den);
  }
}

// ---------------------------------------------------------------------------
// Fused attention + output-projection (B=1 decode, max_seq <= 4096).
//
// Why fuse: the proj GEMV's weight stream (M*K bf16, e.g. 33.6 MB for
// Llama-3-8B) has NO data dependency on the attention output — only the
// dot phase does.  As two launches, the stream cannot start until
// attention finishes, so the layer pays attention latency (~10 us, the
// 8-block kernel is latency-bound at decode S) PLUS the proj kernel's
// cold-start ramp (~11 us measured: profiles/decode_token_profile_r01.md).
// In ONE launch, blocks [0, n_kv) run the attention (identical to
// attn_decode_block_kernel) and publish y with write-through 4-byte
// agent-scope stores + ONE tag-valued flag word per kv head (guide
// Guideline 16 R1: sc1 payload, vmcnt(0) drain per storing wave, relaxed
// flag); blocks [n_kv, n_kv+PB) stage their proj W rows into LDS at full
// HBM rate MEANWHILE, then ONE LANE polls the n_kv flag words, one
// agent-scope acquire fence covers the block, and y is bulk-loaded with
// plain coalesced reads.  (A first version swept 8-byte data-tagged
// granules from every consumer block — measured 285 vs 320 tok/s on
// 8B: 512 sweepers re-reading 16 KB all phase starve the weight stream,
// the price table's polling-cost row.  One lane polling n_kv words is
// ~1000x less poll traffic.)
//
// Tagging: tag = slot*max_seq + (pos+1) — unique per (slot, position),
// never 0, identical across a hipGraph replay's captured arguments (the
// kernel reads pos/slot from device scalars), so the scratch needs
// zeroing only at engine reset, not per launch.
// Spin bound: consumers give up after ~5e6 polls and poison y with NaN
// (loud wrong, never a hung GPU).
//
// Per-layer scratch layout (int32): [K/2 y-pairs][16 flag slots].
// ---------------------------------------------------------------------------
typedef __attribute__((address_space(1))) unsigned int gu32_t;

template <int QPK, int HS, int THROTTLE>
__global__ void attn_proj_kernel(
    bf16* __restrict__ out,       // [M] = res + bias + W @ y
    const bf16* __restrict__ qkv, // [qkv_dim] interleaved, RAW
    bf16* __restrict__ kpool, bf16* __restrict__ vpool,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    int rope_ne, const int* __restrict__ pos_p,
    const int* __restrict__ slot_p, int layer, int n_layers_pool,
    int n_kv_heads, int max_seq, float scale,
    const bf16* __restrict__ W,    // [M, K] proj weight, K = n_head*HS
    const bf16* __restrict__ bias, // [M] or null
    const bf16* __restrict__ res,  // [M] residual (x)
    unsigned int* __restrict__ gran,  // [K/2 + 16] per-layer y + flags
    int M, int BR) {
  constexpr int head_size = HS;
  constexpr int ODIM = HS * QPK / 64;
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const int K = n_kv_heads * QPK * head_size;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int S = pos_p[0] + 1;
  const int pos = S - 1;
  const int slot = slot_p[0];
  const unsigned tag = (unsigned)(slot * max_seq + S);
  gu32_t* g32 = (gu32_t*)gran;
  gu32_t* flags = g32 + K / 2;  // [n_kv_heads] tag-valued flag words

  if ((int)blockIdx.x >= n_kv_heads) {
    // ------------------------- projection role -------------------------
    // Grid-stride over row groups: the grid is capped at full residency
    // (2 blocks/CU: 8 waves/CU is the hard wave limit), so no block ever
    // queues behind a first wave of blocks — a non-resident straggler
    // staging its tile alone after everyone else cost ~6 us/layer in the
    // first version of this kernel.
    const int pb = (int)blockIdx.x - n_kv_heads;
    const int PBe = (int)gridDim.x - n_kv_heads;
    bf16* wlds = reinterpret_cast<bf16*>(smem);            // [BR, K]
    int* vflag = reinterpret_cast<int*>(smem + (size_t)BR * K * 2);
    volatile int* yready = vflag + 1;
    const bf16* ygl = reinterpret_cast<const bf16*>(gran); // y after fence
    bool polled = false;
    bool y_ok = true;
    // temporal throttle (MDI_ATTN_PROJ_SLEEP s_sleep units between
    // 64-B/lane staging batches while attention is still running): the
    // PB sweep showed the attention stretch is GLOBAL memory pressure,
    // so the staging rate — not placement — is the lever
    if (tid == 0) *yready = 0;
    __syncthreads();
    for (int row0 = pb * BR; row0 < M; row0 += PBe * BR) {
      if (row0 != pb * BR) __syncthreads();  // prev dot read wlds
      const int nr = min(BR, M - row0);
      // all 4 waves stage nr rows of W (nt: streamed once).  4 loads in
      // flight per lane: a serial load->ds_write chain is latency-bound
      // (one outstanding 16 B load/lane ~ 2 TB/s chip-wide, the very
      // ramp this kernel exists to remove)
      {
        const int chunks = nr * (K / 8);
        int i = tid;
        for (; i + 3 * 256 < chunks; i += 4 * 256) {
          if (THROTTLE > 0 && !polled && !*yready) {
            if (lane == 0) {
              const unsigned f = __hip_atomic_load(
                  &flags[wave & (n_kv_heads - 1)], __ATOMIC_RELAXED,
                  __HIP_MEMORY_SCOPE_AGENT);
              if (f == tag) *yready = 1;
            }
            __builtin_amdgcn_s_sleep(THROTTLE);
          }
          bf16x8 w0, w1, w2, w3;
#pragma unroll
          for (int u = 0; u < 4; ++u) {
            const int ii = i + u * 256;
            const int r = ii / (K / 8);
            const int c = ii - r * (K / 8);
            bf16x8 wv = load8_nt(W + (size_t)(row0 + r) * K + c * 8);
            if (u == 0) w0 = wv;
            else if (u == 1) w1 = wv;
            else if (u == 2) w2 = wv;
            else w3 = wv;
          }
#pragma unroll
          for (int u = 0; u < 4; ++u) {
            const int ii = i + u * 256;
            const int r = ii / (K / 8);
            const int c = ii - r * (K / 8);
            const bf16x8& wv = u == 0 ? w0 : u == 1 ? w1 : u == 2 ? w2 : w3;
            *reinterpret_cast<int4*>(wlds + (size_t)r * K + c * 8) =
                *reinterpret_cast<const int4*>(wv.v);
          }
        }
        for (; i < chunks; i += 256) {
          const int r = i / (K / 8);
          const int c = i - r * (K / 8);
          bf16x8 wv = load8_nt(W + (size_t)(row0 + r) * K + c * 8);
          *reinterpret_cast<int4*>(wlds + (size_t)r * K + c * 8) =
              *reinterpret_cast<int4*>(wv.v);
        }
      }
      if (!polled && tid == 0) {
        // ONE lane polls the n_kv flag words; one agent-scope acquire
        // covers the block after the barrier (guide Guideline 16)
        unsigned spins = 0;
        bool ok = true;
        for (int h = 0; h < n_kv_heads; ++h) {
          while (__hip_atomic_load(&flags[h], __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT) != tag) {
            if (++spins > 5000000u) { ok = false; break; }
            __builtin_amdgcn_s_sleep(2);
          }
          if (!ok) break;
        }
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        *vflag = ok ? 1 : 0;
      }
      __syncthreads();
      if (!polled) {
        y_ok = *vflag != 0;
        polled = true;
      }
      // dot phase: wave w computes rows r = w, w+4, ... from LDS W and
      // L2-hot y (8-16 KB, read by every block after the acquire)
      for (int r = wave; r < nr; r += 4) {
        float acc = 0.f;
        for (int i = lane * 8; i < K; i += 64 * 8) {
          bf16x8 wv = load8(wlds + (size_t)r * K + i);
          bf16x8 yv = load8(ygl + i);
#pragma unroll
          for (int j = 0; j < 8; ++j) acc += b2f(wv.v[j]) * b2f(yv.v[j]);
        }
        acc = wave_reduce_sum(acc);
        if (lane == 0) {
          const int grow = row0 + r;
          float a = y_ok ? acc : __int_as_float(0x7FC00000);  // poison
          if (bias != nullptr) a += b2f(bias[grow]);
          if (res != nullptr) a += b2f(res[grow]);
          out[grow] = f2b(a);
        }
      }
    }
    return;
  }

  // --------------------------- attention role ---------------------------
  // identical computation to attn_decode_block_kernel<QPK, HS, 0>, with
  // the combine epilogue publishing granules instead of storing y
  bf16* q_lds = reinterpret_cast<bf16*>(smem);      // [16 * HS] (swizzled)
  bf16* k_cur = q_lds + 16 * head_size;             // [HS]
  float* o_part = reinterpret_cast<float*>(
      smem + ((17 * head_size * 2 + 15) & ~15));    // [4][QPK][HS]
  float* ml_part = o_part + ATTN_WAVES * QPK * head_size;  // [4][QPK][2]

  const int g = (int)blockIdx.x;

  const size_t cache_base =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
      (size_t)max_seq * head_size;
  const bf16* krow_cur = qkv + ((size_t)g * (QPK + 2) + QPK) * head_size;
  const bf16* vrow_cur = krow_cur + head_size;

  for (int i = tid; i < (16 - QPK) * head_size / 8; i += 256)
    reinterpret_cast<bf16x8_t*>(q_lds + QPK * head_size)[i] = bf16x8_t{};
  for (int i = tid; i < QPK * head_size; i += 256) {
    const int r = i / head_size;
    const int d = i % head_size;
    const bf16* qrow = qkv + ((size_t)g * (QPK + 2) + r) * head_size;
    q_lds[q_swz<HS>(r, d)] =
        f2b(rope_elem(qrow, d, rope_ne, cos_t, sin_t, pos));
  }
  for (int d = tid; d < head_size; d += 256)
    k_cur[d] = f2b(rope_elem(krow_cur, d, rope_ne, cos_t, sin_t, pos));
  __syncthreads();

  for (int d = tid; d < head_size; d += 256) {
    kpool[cache_base + (size_t)pos * head_size + d] = k_cur[d];
    vpool[cache_base + (size_t)pos * head_size + d] = vrow_cur[d];
  }

  const int qb = lane % QPK;
  const int sub = lane >> 4;
  const int d0 = (lane / QPK) * ODIM;

  float m_st = -1e30f, l_st = 0.f;
  float o_acc[ODIM];
#pragma unroll
  for (int i = 0; i < ODIM; ++i) o_acc[i] = 0.f;

  const int n_tiles = (S + 15) / 16;
  for (int t = wave; t < n_tiles; t += ATTN_WAVES) {
    const int key0 = t * 16;
    f32x4 acc4 = {0.f, 0.f, 0.f, 0.f};
    const int arow = lane & 15;
    const int koff = (lane >> 4) * 8;
    const int akey = key0 + arow;
    const bf16* krow = kpool + cache_base + (size_t)akey * head_size;
    const bool row_valid = akey < S;
    const bool row_cur = akey == pos;
#pragma unroll
    for (int c = 0; c < HS / 32; ++c) {
      bf16x8_t af = {};
      if (row_cur)
        af = *reinterpret_cast<const bf16x8_t*>(&k_cur[c * 32 + koff]);
      else if (row_valid)
        af = *reinterpret_cast<const bf16x8_t*>(krow + c * 32 + koff);
      const bf16x8_t bq = *reinterpret_cast<const bf16x8_t*>(
          &q_lds[q_swz<HS>(arow, c * 32 + koff)]);
      acc4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bq, acc4, 0, 0, 0);
    }
    float sc[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = key0 + sub * 4 + r;
      sc[r] = (key < S) ? acc4[r] * scale : -1e30f;
    }
    float tmax = fmaxf(fmaxf(sc[0], sc[1]), fmaxf(sc[2], sc[3]));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float m_new = fmaxf(m_st, tmax);
    const float alpha = __expf(m_st - m_new);
    m_st = m_new;
    float e[4];
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = key0 + sub * 4 + r;
      e[r] = (key < S) ? __expf(sc[r] - m_new) : 0.f;
      psum += e[r];
    }
    psum += __shfl_xor(psum, 16, 64);
    psum += __shfl_xor(psum, 32, 64);
    l_st = l_st * alpha + psum;
    const float alphaB = __shfl(alpha, qb, 64);
#pragma unroll
    for (int i = 0; i < ODIM; ++i) o_acc[i] *= alphaB;
#pragma unroll
    for (int sI = 0; sI < 16; ++sI) {
      const float pv = __shfl(e[sI & 3], ((sI >> 2) << 4) | qb, 64);
      const int skey = min(key0 + sI, pos);
      const bf16* vrow =
          (skey == pos)
              ? vrow_cur + d0
              : vpool + cache_base + (size_t)skey * head_size + d0;
      if constexpr (ODIM >= 8) {
#pragma unroll
        for (int i = 0; i < ODIM; i += 8) {
          bf16x8 vv = load8(vrow + i);
#pragma unroll
          for (int j = 0; j < 8; ++j) o_acc[i + j] += pv * b2f(vv.v[j]);
        }
      } else if constexpr (ODIM == 4) {
        int2 raw = *reinterpret_cast<const int2*>(vrow);
        const bf16* vv = reinterpret_cast<const bf16*>(&raw);
#pragma unroll
        for (int j = 0; j < 4; ++j) o_acc[j] += pv * b2f(vv[j]);
      } else if constexpr (ODIM == 2) {
        int raw = *reinterpret_cast<const int*>(vrow);
        const bf16* vv = reinterpret_cast<const bf16*>(&raw);
        o_acc[0] += pv * b2f(vv[0]);
        o_acc[1] += pv * b2f(vv[1]);
      } else {
        o_acc[0] += pv * b2f(vrow[0]);
      }
    }
  }

#pragma unroll
  for (int i = 0; i < ODIM; ++i)
    o_part[(wave * QPK + qb) * head_size + d0 + i] = o_acc[i];
  if (lane < QPK) {
    ml_part[(wave * QPK + lane) * 2 + 0] = m_st;
    ml_part[(wave * QPK + lane) * 2 + 1] = l_st;
  }
  __syncthreads();

  // combine + publish: write-through 4-byte pair stores (R1 payload),
  // per-wave vmcnt(0) drain, then ONE lane stores this head's flag
  for (int i = tid; i < QPK * head_size / 2; i += 256) {
    const int e0 = 2 * i;
    const int h = e0 / head_size;
    const int d = e0 % head_size;
    float M_ = -1e30f;
#pragma unroll
    for (int w = 0; w < ATTN_WAVES; ++w)
      M_ = fmaxf(M_, ml_part[(w * QPK + h) * 2 + 0]);
    float den = 0.f, num0 = 0.f, num1 = 0.f;
#pragma unroll
    for (int w = 0; w < ATTN_WAVES; ++w) {
      const float wgt = __expf(ml_part[(w * QPK + h) * 2 + 0] - M_);
      den += wgt * ml_part[(w * QPK + h) * 2 + 1];
      num0 += wgt * o_part[(w * QPK + h) * head_size + d];
      num1 += wgt * o_part[(w * QPK + h) * head_size + d + 1];
    }
    const bf16 y0 = f2b(num0 / den);
    const bf16 y1 = f2b(num1 / den);
    const unsigned pack =
        (unsigned)*reinterpret_cast<const unsigned short*>(&y0) |
        ((unsigned)*reinterpret_cast<const unsigned short*>(&y1) << 16);
    const int gi = (g * QPK * head_size) / 2 + i;
    __hip_atomic_store(&g32[gi], pack, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
  }
  // EVERY storing wave drains its write-through stores before the flag
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (tid == 0)
    __hip_atomic_store(&flags[g], tag, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
}

// BATCH: 0 -> single-token mode (pos_p/slot_p are device scalars);
//        1 -> batched mode: pos_p/slot_p are arrays [n_batch], and
//             qkv/part_o/part_ml/out get a leading batch dimension.
template <int QPK, int HS, int BATCH, int KV8>
__global__ void attn_decode_kernel(
    float* __restrict__ part_o,   // [B?, n_head, n_chunks, head_size]
    float* __restrict__ part_ml,  // [B?, n_head, n_chunks, 2]
    const bf16* __restrict__ qkv, // [B?, qkv_dim] interleaved, RAW
    void* __restrict__ kpool_v, void* __restrict__ vpool_v,
    float* __restrict__ kscale, float* __restrict__ vscale,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    int rope_ne, const int* __restrict__ pos_p,
    const int* __restrict__ slot_p, int layer, int n_layers_pool,
    int n_kv_heads, int max_seq, int n_chunks, float scale, int n_batch) {
  using kvt = std::conditional_t<KV8 != 0, u8kv, bf16>;
  kvt* kpool = reinterpret_cast<kvt*>(kpool_v);
  kvt* vpool = reinterpret_cast<kvt*>(vpool_v);
  constexpr int head_size = HS;
  // q tile + current-token k row are BLOCK-shared: n_chunks is a
  // multiple of ATTN_WAVES, so all 4 waves of a block carry the same
  // (batch, kv-head) and the roped q staging runs once per block, not
  // once per wave (the per-wave form paid ~2-4 us of scalar rope loads
  // per wave — the dominant cost at 256 chunks).  p/m/l/alpha stay
  // per-wave (per-chunk online-softmax state).
  __shared__ __attribute__((aligned(16))) bf16 q_lds[16 * HS];
  __shared__ __attribute__((aligned(16))) bf16 k_cur[HS];
  __shared__ float p_lds[ATTN_WAVES][16][16];
  __shared__ float m_lds[ATTN_WAVES][16];
  __shared__ float l_lds[ATTN_WAVES][16];
  __shared__ float a_lds[ATTN_WAVES][16];

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wg_id = blockIdx.x * ATTN_WAVES + wave;
  const int per_b = n_kv_heads * n_chunks;
  const int b = BATCH ? wg_id / per_b : 0;
  const int rem = BATCH ? wg_id % per_b : wg_id;
  const int g = rem / n_chunks;          // kv head
  const int chunk = rem % n_chunks;
  if (g >= n_kv_heads || b >= n_batch)
    return;  // tail waves exit before any barrier

  const int S = pos_p[b] + 1;  // keys visible this step
  const int pos = S - 1;
  const int slot = slot_p[b];
  const int keys_per_chunk = ((S + n_chunks - 1) / n_chunks + 15) & ~15;
  const int k_begin = chunk * keys_per_chunk;
  // whole-block early exit: this block's 4 chunks are all past S (the
  // combine only reads the first ceil(S / keys_per_chunk) chunks)
  const int c0 = chunk - wave;  // block's first chunk (uniform)
  if ((long long)c0 * keys_per_chunk >= (long long)S) return;
  const int n_head_all = n_kv_heads * QPK;
  if (BATCH) {
    const int qkv_dim = n_kv_heads * (QPK + 2) * head_size;
    qkv += (size_t)b * qkv_dim;
    part_o += (size_t)b * n_head_all * n_chunks * head_size;
    part_ml += (size_t)b * n_head_all * n_chunks * 2;
  }

  const size_t cache_base =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
      (size_t)max_seq * head_size;
  const size_t scl_base =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
      (size_t)max_seq;
  const bf16* krow_cur = qkv + ((size_t)g * (QPK + 2) + QPK) * head_size;
  const bf16* vrow_cur = krow_cur + head_size;

  // ---- stage q rows (QPK real, rest zero) into LDS, rope fused ----------
  // block-cooperative: 256 threads, once per block
  for (int i = threadIdx.x; i < 16 * head_size; i += 256) {
    const int r = i / head_size;
    const int d = i % head_size;
    bf16 val = f2b(0.f);
    if (r < QPK) {
      const bf16* qrow = qkv + ((size_t)g * (QPK + 2) + r) * head_size;
      val = f2b(rope_elem(qrow, d, rope_ne, cos_t, sin_t, pos));
    }
    q_lds[q_swz<HS>(r, d)] = val;
  }
  // current-token k row, roped (the pool does not hold it yet)
  for (int d = threadIdx.x; d < head_size; d += 256)
    k_cur[d] = f2b(rope_elem(krow_cur, d, rope_ne, cos_t, sin_t, pos));
  if (lane < 16) {
    m_lds[wave][lane] = -1e30f;
    l_lds[wave][lane] = 0.f;
    a_lds[wave][lane] = 1.f;
  }
  __syncthreads();

  // ---- per-lane roles ----------------------------------------------------
  // softmax map (A): qhead qa = lane & 15, key sub-row = lane >> 4
  // PV map (B): qb = lane % QPK, dim slice d0, ODIM dims per lane
  static_assert(HS % 32 == 0 && HS * QPK >= 64, "unsupported attn geometry");
  constexpr int ODIM = HS * QPK / 64;  // output dims per lane
  const int qa = lane & 15;
  const int sub = lane >> 4;
  const int qb = lane % QPK;
  const int d0 = (lane / QPK) * ODIM;

  float o_acc[ODIM];
#pragma unroll
  for (int i = 0; i < ODIM; ++i) o_acc[i] = 0.f;

  const int n_tiles = keys_per_chunk / 16;
  const int arow = lane & 15;          // key row in tile (and B's qhead col)
  const int koff = (lane >> 4) * 8;    // dim offset within 32-chunk

  // K-tile fragments double-buffered in registers: tile t+1's four
  // 16-B loads are issued before tile t's MFMA/softmax/PV consume them
  // (global loads stay in flight across the barriers), so the 2-4 serial
  // tiles per wave at long S no longer stack their load latencies.
  auto load_af = [&](int t, bf16x8_t (&af)[HS / 32]) {
    const int akey = k_begin + t * 16 + arow;
    const kvt* krow = kpool + cache_base + (size_t)akey * head_size;
    const bool row_valid = akey < S;
    const bool row_cur = akey == pos;  // newest key: read from LDS k_cur
    float kscl = 1.f;
    if (KV8 && row_valid && !row_cur) kscl = kscale[scl_base + akey];
#pragma unroll
    for (int c = 0; c < HS / 32; ++c) {
      af[c] = bf16x8_t{};
      if (row_cur) {
        af[c] = *reinterpret_cast<const bf16x8_t*>(&k_cur[c * 32 + koff]);
      } else if (row_valid) {
        if constexpr (KV8)
          af[c] = fp8x8_to_bf16(
              reinterpret_cast<const u8kv*>(krow) + c * 32 + koff, kscl);
        else
          af[c] = *reinterpret_cast<const bf16x8_t*>(
              reinterpret_cast<const bf16*>(krow) + c * 32 + koff);
      }
    }
  };

  auto tile_body = [&](int t, const bf16x8_t (&afr)[HS / 32]) {
    const int key0 = k_begin + t * 16;
    // NOTE: no early break — every wave in the block must reach every
    // __syncthreads(); empty tiles run fully masked.
    f32x4 acc4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int c = 0; c < HS / 32; ++c) {
      const bf16x8_t bf = *reinterpret_cast<const bf16x8_t*>(
          &q_lds[q_swz<HS>(arow, c * 32 + koff)]);
      acc4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr[c], bf, acc4,
                                                     0, 0, 0);
    }
    // C[row=key, col=qhead]: lane holds keys (sub*4 + r) for qhead qa
    float sc[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = key0 + sub * 4 + r;
      sc[r] = (key < S) ? acc4[r] * scale : -1e30f;
    }
    // ---- online softmax per qhead (4 lanes per qhead: strides 16,32) -----
    float tmax = fmaxf(fmaxf(sc[0], sc[1]), fmaxf(sc[2], sc[3]));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    __syncthreads();  // a_lds/p_lds from previous tile fully consumed
    if (sub == 0) {
      float m_old = m_lds[wave][qa];
      float m_new = fmaxf(m_old, tmax);
      a_lds[wave][qa] = __expf(m_old - m_new);
      m_lds[wave][qa] = m_new;
    }
    __syncthreads();
    const float m_new = m_lds[wave][qa];
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float p = __expf(sc[r] - m_new);
      const int key = key0 + sub * 4 + r;
      if (key >= S) p = 0.f;
      p_lds[wave][sub * 4 + r][qa] = p;
      psum += p;
    }
    psum += __shfl_xor(psum, 16, 64);
    psum += __shfl_xor(psum, 32, 64);
    if (sub == 0)
      l_lds[wave][qa] = l_lds[wave][qa] * a_lds[wave][qa] + psum;
    __syncthreads();

    // ---- P*V: o[qb][d0..] += p[s][qb] * V[s][d0..] -----------------------
    const float alphaB = a_lds[wave][qb];
#pragma unroll
    for (int i = 0; i < ODIM; ++i) o_acc[i] *= alphaB;
    const int smax = min(16, S - key0);
    for (int s = 0; s < smax; ++s) {
      const float p = p_lds[wave][s][qb];
      const int skey = key0 + s;
      const bool vcur = skey == pos;
      const bf16* vrow_b = vrow_cur + d0;
      const kvt* vrow_p = vpool + cache_base + (size_t)skey * head_size + d0;
      float vscl = 1.f;
      if (KV8 && !vcur) vscl = vscale[scl_base + skey];
      if constexpr (ODIM >= 8) {
#pragma unroll
        for (int i = 0; i < ODIM; i += 8) {
          if (!vcur) {
            if constexpr (KV8) {
              fp8x8_fma(&o_acc[i], p,
                        reinterpret_cast<const u8kv*>(vrow_p) + i, vscl);
              continue;
            }
          }
          bf16x8_t vv;
          if (vcur) {
            vv = *reinterpret_cast<const bf16x8_t*>(vrow_b + i);
          } else {
            vv = *reinterpret_cast<const bf16x8_t*>(
                reinterpret_cast<const bf16*>(vrow_p) + i);
          }
#pragma unroll
          for (int j = 0; j < 8; ++j) o_acc[i + j] += p * (float)vv[j];
        }
      } else {
#pragma unroll
        for (int j = 0; j < ODIM; ++j) {
          float v;
          if (vcur)
            v = b2f(vrow_b[j]);
          else if constexpr (KV8)
            v = fp8_to_f32(reinterpret_cast<const u8kv*>(vrow_p)[j]) * vscl;
          else
            v = b2f(reinterpret_cast<const bf16*>(vrow_p)[j]);
          o_acc[j] += p * v;
        }
      }
    }
  };

  bf16x8_t afA[HS / 32], afB[HS / 32];
  if (n_tiles > 0) load_af(0, afA);
  for (int t = 0; t < n_tiles; t += 2) {
    if (t + 1 < n_tiles) load_af(t + 1, afB);
    tile_body(t, afA);
    if (t + 2 < n_tiles) load_af(t + 2, afA);
    if (t + 1 < n_tiles) tile_body(t + 1, afB);
  }

  // ---- append the current token's k,v to the pool ------------------------
  // exactly one chunk's range contains pos; its wave owns the append (no
  // other wave reads pool row pos this step — they use k_cur/vrow_cur)
  if (k_begin <= pos && pos < k_begin + keys_per_chunk) {
    if constexpr (KV8) {
      float lk = 0.f, lv = 0.f;
      for (int d = lane; d < head_size; d += 64) {
        lk = fmaxf(lk, fabsf(b2f(k_cur[d])));
        lv = fmaxf(lv, fabsf(b2f(vrow_cur[d])));
      }
      const float ks = fmaxf(wave_reduce_max(lk), 1e-12f) / 448.f;
      const float vs = fmaxf(wave_reduce_max(lv), 1e-12f) / 448.f;
      for (int d = lane; d < head_size; d += 64) {
        kpool[cache_base + (size_t)pos * head_size + d] =
            f32_to_fp8(b2f(k_cur[d]) / ks);
        vpool[cache_base + (size_t)pos * head_size + d] =
            f32_to_fp8(b2f(vrow_cur[d]) / vs);
      }
      if (lane == 0) {
        kscale[scl_base + pos] = ks;
        vscale[scl_base + pos] = vs;
      }
    } else {
      for (int d = lane; d < head_size; d += 64) {
        kpool[cache_base + (size_t)pos * head_size + d] = k_cur[d];
        vpool[cache_base + (size_t)pos * head_size + d] = vrow_cur[d];
      }
    }
  }

  // ---- write partials ----------------------------------------------------
  // global qhead index for PV map: h = g*QPK + qb
  if (k_begin < S) {
    const int hB = g * QPK + qb;
    float* op = part_o + ((size_t)hB * n_chunks + chunk) * head_size + d0;
#pragma unroll
    for (int i = 0; i < ODIM; ++i) op[i] = o_acc[i];
    if (sub == 0 && qa < QPK) {
      const int hA = g * QPK + qa;
      part_ml[((size_t)hA * n_chunks + chunk) * 2 + 0] = m_lds[wave][qa];
      part_ml[((size_t)hA * n_chunks + chunk) * 2 + 1] = l_lds[wave][qa];
    }
  }
  // chunks with k_begin >= S write nothing: the S-aware combine reads
  // only the first ceil(S / keys_per_chunk) chunk slots
}

// Combine split-S partials: out[h][d] = sum_c w_c * part_o[h][c][d] / L.
// One BLOCK per (head, 64-dim slice); each of the 4 waves reduces a
// 64-chunk strip (lane <-> dim, coalesced 256-B loads per chunk, unroll-8
// ILP), partials summed through LDS.  The previous one-wave-per-(h,d)
// form walked all 256 chunks serially from 64 waves total: ~45 us/layer
// at S >= 4k, 10x the split kernel itself.  Only the first
// ceil(S / keys_per_chunk) chunk slots are read (the split kernel writes
// nothing for later chunks).
__global__ void attn_combine_kernel(bf16* __restrict__ out,
                                    const float* __restrict__ part_o,
                                    const float* __restrict__ part_ml,
                                    int n_chunks, int head_size, int n_head,
                                    const int* __restrict__ pos_p,
                                    int n_head_per_b) {
  const int DS = (head_size + 63) / 64;
  const int h = (int)blockIdx.x / DS;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int d = ((int)blockIdx.x % DS) * 64 + lane;
  if (h >= n_head) return;
  const int S = pos_p[h / n_head_per_b] + 1;
  const int kpc = ((S + n_chunks - 1) / n_chunks + 15) & ~15;
  int n_act = (S + kpc - 1) / kpc;
  if (n_act > n_chunks) n_act = n_chunks;

  __shared__ float partial[ATTN_WAVES][64];

  // every wave computes the identical chunk weights (lane c holds chunks
  // c, c+64, c+128, c+192 in four registers)
  float mr[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float lr[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int c = lane + r * 64;
    if (c < n_act) {
      mr[r] = part_ml[((size_t)h * n_chunks + c) * 2];
      lr[r] = part_ml[((size_t)h * n_chunks + c) * 2 + 1];
    }
  }
  float M = fmaxf(fmaxf(mr[0], mr[1]), fmaxf(mr[2], mr[3]));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    M = fmaxf(M, __shfl_xor(M, off, 64));
  float wr[4];
  float wl = 0.f;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    wr[r] = (lane + r * 64 < n_act) ? __expf(mr[r] - M) : 0.f;
    wl += wr[r] * lr[r];
  }
  const float inv = 1.f / wave_reduce_sum(wl);

  // wave w reduces chunks [w*64, w*64+64) for its dim
  const float wrw = wave == 0 ? wr[0] : wave == 1 ? wr[1]
                  : wave == 2 ? wr[2] : wr[3];
  float acc = 0.f;
  if (d < head_size) {
    const int c0 = wave * 64;
    const int ce = n_act < c0 + 64 ? n_act : c0 + 64;
    const float* po = part_o + (size_t)h * n_chunks * head_size + d;
#pragma unroll 8
    for (int c = c0; c < ce; ++c)
      acc += __shfl(wrw, c - c0, 64) * po[(size_t)c * head_size];
  }
  partial[wave][lane] = acc;
  __syncthreads();
  if (wave == 0 && d < head_size) {
    const float a = partial[0][lane] + partial[1][lane] +
                    partial[2][lane] + partial[3][lane];
    out[(size_t)h * head_size + d] = f2b(a * inv);
  }
}

// ---------------------------------------------------------------------------
// Prefill: rope q/k for ALL T prompt positions + append k/v to the pool.
// qkv: [T, qkv_dim] (roped in place); grid (n_kv_heads, T).
// ---------------------------------------------------------------------------
template <int KV8>
__global__ void rope_prefill_append_kernel(
    bf16* __restrict__ qkv, void* __restrict__ kpool_v,
    void* __restrict__ vpool_v, float* __restrict__ kscale,
    float* __restrict__ vscale, const float* __restrict__ cos_t,
    const float* __restrict__ sin_t, int pos0, int slot, int layer,
    int n_layers_pool, int n_kv_heads, int max_seq, int head_size,
    int rope_ne, int qpk) {
  using kvt = std::conditional_t<KV8 != 0, u8kv, bf16>;
  kvt* kpool = reinterpret_cast<kvt*>(kpool_v);
  kvt* vpool = reinterpret_cast<kvt*>(vpool_v);
  __shared__ unsigned kvmax[2];
  const int g = blockIdx.x;
  const int t = blockIdx.y;
  const int pos = pos0 + t;
  const int tid = threadIdx.x;
  const int half = rope_ne >> 1;
  const int qkv_dim = n_kv_heads * (qpk + 2) * head_size;

  bf16* base = qkv + (size_t)t * qkv_dim + (size_t)g * (qpk + 2) * head_size;
  // rope q rows + k row: process pairs (d, d+half)
  for (int idx = tid; idx < (qpk + 1) * half; idx += blockDim.x) {
    const int r = idx / half;
    const int d = idx % half;
    bf16* row = base + (size_t)r * head_size;
    const float x1 = b2f(row[d]);
    const float x2 = b2f(row[d + half]);
    const float c1 = cos_t[(size_t)pos * rope_ne + d];
    const float s1 = sin_t[(size_t)pos * rope_ne + d];
    const float c2 = cos_t[(size_t)pos * rope_ne + d + half];
    const float s2 = sin_t[(size_t)pos * rope_ne + d + half];
    row[d] = f2b(x1 * c1 - x2 * s1);
    row[d + half] = f2b(x2 * c2 + x1 * s2);
  }
  if (KV8 && tid < 2) kvmax[tid] = 0;
  __syncthreads();
  const size_t cache_off =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
          (size_t)max_seq * head_size +
      (size_t)pos * head_size;
  const bf16* krow = base + (size_t)qpk * head_size;
  const bf16* vrow = base + (size_t)(qpk + 1) * head_size;
  if constexpr (KV8) {
    float lk = 0.f, lv = 0.f;
    for (int d = tid; d < head_size; d += blockDim.x) {
      lk = fmaxf(lk, fabsf(b2f(krow[d])));
      lv = fmaxf(lv, fabsf(b2f(vrow[d])));
    }
    lds_fmax_u(&kvmax[0], lk);
    lds_fmax_u(&kvmax[1], lv);
    __syncthreads();
    const float ks = fmaxf(__uint_as_float(kvmax[0]), 1e-12f) / 448.f;
    const float vs = fmaxf(__uint_as_float(kvmax[1]), 1e-12f) / 448.f;
    for (int d = tid; d < head_size; d += blockDim.x) {
      kpool[cache_off + d] = f32_to_fp8(b2f(krow[d]) / ks);
      vpool[cache_off + d] = f32_to_fp8(b2f(vrow[d]) / vs);
    }
    if (tid == 0) {
      const size_t sb =
          (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
              (size_t)max_seq +
          (size_t)pos;
      kscale[sb] = ks;
      vscale[sb] = vs;
    }
  } else {
    for (int d = tid; d < head_size; d += blockDim.x) {
      kpool[cache_off + d] = krow[d];
      vpool[cache_off + d] = vrow[d];
    }
  }
}

// ---------------------------------------------------------------------------
// Prefill flash attention (causal, GQA): MFMA QK^T with online softmax.
//
// One WAVE handles one (query head h, 16-query-row tile): Q tile staged in
// LDS (roped already), K/V streamed from the pool in 16-key tiles, causal
// per-row masking, online (m,l) rescaling, O normalized and written
// directly — no S^2 score materialization, no split-S partials.
// grid: ceil(n_head * n_qtiles / 4) blocks of 256 threads.
// ---------------------------------------------------------------------------
template <int HS, int KV8>
__global__ void prefill_attn_kernel(
    bf16* __restrict__ out,        // [T, n_head*HS]
    const bf16* __restrict__ qkv,  // [T, qkv_dim], q already roped
    const void* __restrict__ kpool_v, const void* __restrict__ vpool_v,
    const float* __restrict__ kscale, const float* __restrict__ vscale,
    int pos0, int slot, int layer, int n_layers_pool, int n_kv_heads,
    int max_seq, int qpk, int T, float scale) {
  using kvt = std::conditional_t<KV8 != 0, u8kv, bf16>;
  const kvt* kpool = reinterpret_cast<const kvt*>(kpool_v);
  const kvt* vpool = reinterpret_cast<const kvt*>(vpool_v);
  __shared__ __attribute__((aligned(16))) bf16 q_lds[ATTN_WAVES][16 * HS];
  __shared__ float p_lds[ATTN_WAVES][16][16];
  __shared__ float m_lds[ATTN_WAVES][16];
  __shared__ float l_lds[ATTN_WAVES][16];
  __shared__ float a_lds[ATTN_WAVES][16];

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n_head = n_kv_heads * qpk;
  const int n_qtiles = (T + 15) / 16;
  const int wg_id = blockIdx.x * ATTN_WAVES + wave;
  const int h = wg_id / n_qtiles;        // query head
  const int qtile = wg_id % n_qtiles;
  if (h >= n_head) return;
  const int g = h / qpk;                 // kv head
  const int hj = h % qpk;                // q row within group
  const int q_base = qtile * 16;         // first q row (relative to prompt)
  const int qkv_dim = n_kv_heads * (qpk + 2) * HS;

  const size_t cache_base =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
      (size_t)max_seq * HS;
  const size_t scl_base =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
      (size_t)max_seq;

  // stage 16 q rows (t = q_base..q_base+15) of head h into LDS
  for (int i = lane; i < 16 * (HS / 8); i += 64) {
    const int r = i / (HS / 8);
    const int d8 = (i % (HS / 8)) * 8;
    int4 val = {0, 0, 0, 0};
    const int t = q_base + r;
    if (t < T) {
      const bf16* qrow =
          qkv + (size_t)t * qkv_dim + ((size_t)g * (qpk + 2) + hj) * HS + d8;
      val = *reinterpret_cast<const int4*>(qrow);
    }
    *reinterpret_cast<int4*>(&q_lds[wave][q_swz<HS>(r, d8)]) = val;
  }
  if (lane < 16) {
    m_lds[wave][lane] = -1e30f;
    l_lds[wave][lane] = 0.f;
    a_lds[wave][lane] = 1.f;
  }
  __syncthreads();

  constexpr int ODIM = HS * 16 / 64;  // dims per lane in the PV map
  const int qa = lane & 15;   // q row (softmax map)
  const int sub = lane >> 4;
  const int qb = lane % 16;   // q row (PV map)
  const int d0 = (lane / 16) * ODIM;

  float o_acc[ODIM];
#pragma unroll
  for (int i = 0; i < ODIM; ++i) o_acc[i] = 0.f;

  // keys visible to the LAST row of this tile (causal upper bound)
  const int k_last = pos0 + min(q_base + 15, T - 1) + 1;
  const int n_tiles = (k_last + 15) / 16;
  for (int kt = 0; kt < n_tiles; ++kt) {
    const int key0 = kt * 16;
    // ---- scores via MFMA: A = K tile, B = Q^T --------------------------
    f32x4 acc4 = {0.f, 0.f, 0.f, 0.f};
    const int arow = lane & 15;
    const int koff = (lane >> 4) * 8;
    const kvt* krow = kpool + cache_base + (size_t)(key0 + arow) * HS;
    const bool row_valid = (key0 + arow) < k_last;
    float kscl = 1.f;
    if (KV8 && row_valid) kscl = kscale[scl_base + key0 + arow];
#pragma unroll
    for (int c = 0; c < HS / 32; ++c) {
      bf16x8_t af = {};
      if (row_valid) {
        if constexpr (KV8)
          af = fp8x8_to_bf16(
              reinterpret_cast<const u8kv*>(krow) + c * 32 + koff, kscl);
        else
          af = *reinterpret_cast<const bf16x8_t*>(
              reinterpret_cast<const bf16*>(krow) + c * 32 + koff);
      }
      const bf16x8_t bfr = *reinterpret_cast<const bf16x8_t*>(
          &q_lds[wave][q_swz<HS>(arow, c * 32 + koff)]);
      acc4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bfr, acc4, 0, 0, 0);
    }
    // causal mask: key (abs) must be <= q position (abs)
    float sc[4];
    const int q_abs = pos0 + q_base + qa;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = key0 + sub * 4 + r;
      sc[r] = (key <= q_abs && key < k_last) ? acc4[r] * scale : -1e30f;
    }
    float tmax = fmaxf(fmaxf(sc[0], sc[1]), fmaxf(sc[2], sc[3]));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    __syncthreads();
    if (sub == 0) {
      const float m_old = m_lds[wave][qa];
      const float m_new = fmaxf(m_old, tmax);
      a_lds[wave][qa] = __expf(m_old - m_new);
      m_lds[wave][qa] = m_new;
    }
    __syncthreads();
    const float m_new = m_lds[wave][qa];
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float p = __expf(sc[r] - m_new);
      const int key = key0 + sub * 4 + r;
      if (key > q_abs || key >= k_last) p = 0.f;
      p_lds[wave][sub * 4 + r][qa] = p;
      psum += p;
    }
    psum += __shfl_xor(psum, 16, 64);
    psum += __shfl_xor(psum, 32, 64);
    if (sub == 0)
      l_lds[wave][qa] = l_lds[wave][qa] * a_lds[wave][qa] + psum;
    __syncthreads();

    const float alphaB = a_lds[wave][qb];
#pragma unroll
    for (int i = 0; i < ODIM; ++i) o_acc[i] *= alphaB;
    const int smax = min(16, k_last - key0);
    for (int s = 0; s < smax; ++s) {
      const float p = p_lds[wave][s][qb];
      const kvt* vrow = vpool + cache_base + (size_t)(key0 + s) * HS + d0;
      float vscl = 1.f;
      if (KV8) vscl = vscale[scl_base + key0 + s];
#pragma unroll
      for (int i = 0; i < ODIM; i += 8) {
        bf16x8_t vv;
        if constexpr (KV8)
          vv = fp8x8_to_bf16(reinterpret_cast<const u8kv*>(vrow) + i,
                             vscl);
        else
          vv = *reinterpret_cast<const bf16x8_t*>(
              reinterpret_cast<const bf16*>(vrow) + i);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (i + j < ODIM) o_acc[i + j] += p * (float)vv[j];
      }
    }
  }

  // normalize + write
  const int t_out = q_base + qb;
  if (t_out < T) {
    const float inv = 1.f / l_lds[wave][qb];
    bf16* op = out + (size_t)t_out * (n_head * HS) + (size_t)h * HS + d0;
#pragma unroll
    for (int i = 0; i < ODIM; ++i) op[i] = f2b(o_acc[i] * inv);
  }
}

// ---------------------------------------------------------------------------
// Prefill flash attention v2 (causal, GQA, all-MFMA): the default path.
//
// Two structural fixes over prefill_attn_kernel above (which remains as the
// MDI_PREFILL_V1=1 fallback):
//   1. The whole WORKGROUP shares one K/V LDS staging per 64-key chunk.  The
//      4 waves are the GQA group (hpb = min(qpk,4) query heads of ONE kv
//      head; leftover wave capacity covers extra q tiles), so K/V cross HBM
//      once per group instead of once per query head.
//   2. The P·V product is MFMA too (the v1 kernel spent ~2048 scalar VALU
//      FMAs per wave per chunk on it — 74% of a T=4096 prefill).  P is
//      round-tripped through a per-wave transposed LDS tile to match the
//      A-fragment layout; V is staged transposed so B-fragments are single
//      conflict-free ds_read_b128s (row strides padded to 16 B multiples
//      that are coprime-ish with the 64 LDS banks).
// Softmax state (m, l) lives in registers, replicated across the 4
// sub-tiles of a wave and exchanged with __shfl — no LDS, no barriers
// beyond the two per-chunk staging fences.
// Reference behavior: /root/reference/src/sub/model.py:738-751 (SDPA with
// causal bool mask at prefill), recomputed with online softmax.
// ---------------------------------------------------------------------------
// q-tile swizzle for the v2 prefill kernel: the b128 read groups mix
// arow 0-3/12-15 (sub 0) with arow 4-11 (sub 1), so the XOR key must
// separate all 16 rows — (row&15)<<4 when the row stride allows (HS>=128)
template <int HS>
DEVINL int q_swz2(int row, int d) {
  int byte = (row * HS + d) * 2;
  byte ^= (row & (HS >= 128 ? 15 : 7)) << 4;
  return byte >> 1;
}

template <int HS, int KV8, int PF_KCH>  // PF_KCH: keys staged per chunk
__global__ __launch_bounds__(256) void prefill_attn_mfma_kernel(
    bf16* __restrict__ out,        // [T, n_head*HS]
    const bf16* __restrict__ qkv,  // [T, qkv_dim], q already roped
    const void* __restrict__ kpool_v, const void* __restrict__ vpool_v,
    const float* __restrict__ kscale, const float* __restrict__ vscale,
    int pos0, int slot, int layer, int n_layers_pool, int n_kv_heads,
    int max_seq, int qpk, int T, float scale, int hpb, int qtpb, int n_hgrp,
    int n_qtg) {
  using kvt = std::conditional_t<KV8 != 0, u8kv, bf16>;
  const kvt* kpool = reinterpret_cast<const kvt*>(kpool_v);
  const kvt* vpool = reinterpret_cast<const kvt*>(vpool_v);
  __shared__ __attribute__((aligned(16))) bf16 k_lds[PF_KCH][HS + 16];
  __shared__ __attribute__((aligned(16))) bf16 v_t[HS][PF_KCH + 24];
  __shared__ __attribute__((aligned(16))) bf16 q_lds[ATTN_WAVES][16 * HS];
  __shared__ __attribute__((aligned(16)))
      bf16 p_t[ATTN_WAVES][16][PF_KCH + 24];

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n_qtiles = (T + 15) / 16;
  int bid = blockIdx.x;
  const int g = bid / (n_hgrp * n_qtg);  // kv head
  bid -= g * (n_hgrp * n_qtg);
  const int hgrp = bid / n_qtg;
  const int qtg = bid % n_qtg;
  const int hj = hgrp * hpb + (wave % hpb);  // q row within the kv group
  const int qtile = qtg * qtpb + (wave / hpb);
  // wave/hpb >= qtpb happens when hpb doesn't divide 4 (hpb=3): that wave
  // would claim a q tile outside the block's staged causal bound — idle it
  const bool active =
      (hj < qpk) && (qtile < n_qtiles) && (wave / hpb < qtpb);
  const int h = g * qpk + hj;            // query head (output numbering)
  const int q_base = qtile * 16;
  const int qkv_dim = n_kv_heads * (qpk + 2) * HS;

  const size_t cache_base =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
      (size_t)max_seq * HS;
  const size_t scl_base =
      (((size_t)slot * n_layers_pool + layer) * n_kv_heads + g) *
      (size_t)max_seq;

  // stage this wave's 16 q rows (zero-filled when inactive / past T)
  for (int i = lane; i < 16 * (HS / 8); i += 64) {
    const int r = i / (HS / 8);
    const int d8 = (i % (HS / 8)) * 8;
    int4 val = {0, 0, 0, 0};
    const int t = q_base + r;
    if (active && t < T)
      val = *reinterpret_cast<const int4*>(
          qkv + (size_t)t * qkv_dim + ((size_t)g * (qpk + 2) + hj) * HS + d8);
    *reinterpret_cast<int4*>(&q_lds[wave][q_swz2<HS>(r, d8)]) = val;
  }

  // causal bounds: block = staging bound (last q row any wave covers),
  // wave = this wave's own tile
  const int qt_last = min(qtg * qtpb + qtpb, n_qtiles) * 16 - 1;
  const int k_last_blk = pos0 + min(qt_last, T - 1) + 1;
  const int k_last_w = active ? (pos0 + min(q_base + 15, T - 1) + 1) : 0;

  const int arow = lane & 15;
  const int sub = lane >> 4;
  const int koff = sub * 8;
  const int q_abs = pos0 + q_base + arow;
  // p_t write/read column swizzle (same rationale as the v_t one)
  const int pkx = ((arow >> 3) & (PF_KCH / 8 - 1)) << 3;

  float m_r = -1e30f, l_r = 0.f;
  f32x4 o_pv[HS / 16];
#pragma unroll
  for (int s = 0; s < HS / 16; ++s) o_pv[s] = {0.f, 0.f, 0.f, 0.f};

  const int n_ch = (k_last_blk + PF_KCH - 1) / PF_KCH;
  for (int ch = 0; ch < n_ch; ++ch) {
    const int key0 = ch * PF_KCH;
    __syncthreads();  // previous chunk's readers done (q staging on ch 0)
    // ---- cooperative K / V^T staging (fp8 dequantized once, here) ----
    for (int i = threadIdx.x; i < PF_KCH * (HS / 8); i += 256) {
      const int r = i / (HS / 8);
      const int d8 = (i % (HS / 8)) * 8;
      const int key = key0 + r;
      bf16x8_t kk = {}, vv = {};
      if (key < k_last_blk) {
        const size_t off = cache_base + (size_t)key * HS + d8;
        if constexpr (KV8) {
          kk = fp8x8_to_bf16(kpool + off, kscale[scl_base + key]);
          vv = fp8x8_to_bf16(vpool + off, vscale[scl_base + key]);
        } else {
          kk = *reinterpret_cast<const bf16x8_t*>(
              reinterpret_cast<const bf16*>(kpool) + off);
          vv = *reinterpret_cast<const bf16x8_t*>(
              reinterpret_cast<const bf16*>(vpool) + off);
        }
      }
      *reinterpret_cast<bf16x8_t*>(&k_lds[r][d8]) = kk;
      // key-column XOR swizzle: v_t rows are 36 dwords apart, so rows 8
      // apart hit one ds_write bank (mod 32) — the per-(row>>3) XOR on
      // the key index spreads the 16-way write conflict to 2-way while
      // keeping each b128 read's 8 keys contiguous (key0 stays a
      // multiple of 8 under the XOR)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int vrow = d8 + j;
        const int kc = r ^ (((vrow >> 3) & (PF_KCH / 8 - 1)) << 3);
        *reinterpret_cast<__bf16*>(&v_t[vrow][kc]) = vv[j];
      }
    }
    __syncthreads();
    if (!active || key0 >= k_last_w) continue;  // barriers are at loop top

    // interior chunks (every key causally visible to every q row of the
    // tile) skip the per-key mask compares — wave-uniform
    const bool full = (key0 + PF_KCH - 1) <= (pos0 + q_base);
    // ---- scores: 4 MFMA 16-key tiles against this wave's q tile ----
    float sc[PF_KCH / 16][4];
#pragma unroll
    for (int t4 = 0; t4 < PF_KCH / 16; ++t4) {
      f32x4 a4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int c = 0; c < HS / 32; ++c) {
        const bf16x8_t af = *reinterpret_cast<const bf16x8_t*>(
            &k_lds[t4 * 16 + arow][c * 32 + koff]);
        const bf16x8_t bfr = *reinterpret_cast<const bf16x8_t*>(
            &q_lds[wave][q_swz2<HS>(arow, c * 32 + koff)]);
        a4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bfr, a4, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int key = key0 + t4 * 16 + sub * 4 + r;
        sc[t4][r] =
            full ? a4[r] * scale
                 : ((key <= q_abs && key < k_last_w) ? a4[r] * scale
                                                     : -1e30f);
      }
    }
    // ---- online softmax, state in registers (per qa = arow) ----
    float tmax = -1e30f;
#pragma unroll
    for (int t4 = 0; t4 < PF_KCH / 16; ++t4)
#pragma unroll
      for (int r = 0; r < 4; ++r) tmax = fmaxf(tmax, sc[t4][r]);
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float m_new = fmaxf(m_r, tmax);
    const float alpha = __expf(m_r - m_new);
    m_r = m_new;
    float psum = 0.f;
#pragma unroll
    for (int t4 = 0; t4 < PF_KCH / 16; ++t4)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = __expf(sc[t4][r] - m_new);  // masked rows: exactly 0
        psum += p;
        p_t[wave][arow][(t4 * 16 + sub * 4 + r) ^ pkx] = f2b(p);
      }
    psum += __shfl_xor(psum, 16, 64);
    psum += __shfl_xor(psum, 32, 64);
    l_r = l_r * alpha + psum;

    // ---- P·V via MFMA (A = P rows from p_t, B = V^T slices) ----
    float al[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) al[r] = __shfl(alpha, sub * 4 + r, 64);
#pragma unroll
    for (int s = 0; s < HS / 16; ++s)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_pv[s][r] *= al[r];
#pragma unroll
    for (int k2 = 0; k2 < PF_KCH / 32; ++k2) {
      const bf16x8_t ap = *reinterpret_cast<const bf16x8_t*>(
          &p_t[wave][arow][(k2 * 32 + koff) ^ pkx]);
#pragma unroll
      for (int s = 0; s < HS / 16; ++s) {
        const int vrow = s * 16 + arow;
        const int kc0 = (k2 * 32 + koff) ^
                        (((vrow >> 3) & (PF_KCH / 8 - 1)) << 3);
        const bf16x8_t vb =
            *reinterpret_cast<const bf16x8_t*>(&v_t[vrow][kc0]);
        o_pv[s] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, vb, o_pv[s],
                                                          0, 0, 0);
      }
    }
  }

  // ---- normalize + write (lane holds q rows sub*4+r, dim s*16+arow) ----
  if (active) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int t_out = q_base + sub * 4 + r;
      const float lq = __shfl(l_r, sub * 4 + r, 64);
      if (t_out >= T) continue;
      const float inv = 1.f / lq;
      bf16* op = out + (size_t)t_out * (n_kv_heads * qpk * HS) +
                 (size_t)h * HS + arow;
#pragma unroll
      for (int s = 0; s < HS / 16; ++s) op[s * 16] = f2b(o_pv[s][r] * inv);
    }
  }
}

// ---------------------------------------------------------------------------
// Residual add: out = a + b (bf16, fp32 math)
// ---------------------------------------------------------------------------
__global__ void add_kernel(bf16* __restrict__ out, const bf16* __restrict__ a,
                           const bf16* __restrict__ b, int n) {
  for (int i = (blockIdx.x * blockDim.x + threadIdx.x) * 8; i < n;
       i += gridDim.x * blockDim.x * 8) {
    bf16x8 va = load8(a + i);
    bf16x8 vb = load8(b + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = f2b(b2f(va.v[j]) + b2f(vb.v[j]));
    *reinterpret_cast<int4*>(out + i) = *reinterpret_cast<const int4*>(o.v);
  }
}

// ---------------------------------------------------------------------------
// Fused sampling: temperature + top-k (exact, radix-select on bf16 bits) +
// Gumbel-max draw.  Replaces the reference's torch sample()
// (/root/reference/src/sub/model.py:34-90) on the decode hot path — the
// torch composition costs ~166 us/token on a 128k vocab; this pipeline of
// 5 small kernels costs ~15 us.  Deterministic given (seed, slot, pos).
//
// scratch layout (int32/u32, >= 520 entries, zeroed before each call):
//   [0..255]   hi-byte histogram
//   [256..511] lo-byte histogram
//   [512] bucket_hi, [513] count_above, [514] threshold_u16
//   [516..517] packed (score,idx) u64 argmax cell (8-byte aligned)
// ---------------------------------------------------------------------------

DEVINL unsigned bf16_sortable(unsigned short bits) {
  return (bits & 0x8000u) ? (unsigned)(~bits & 0xFFFFu)
                          : (unsigned)(bits | 0x8000u);
}

DEVINL unsigned hash_u32(unsigned x) {
  x ^= x >> 16;
  x *= 0x7feb352du;
  x ^= x >> 15;
  x *= 0x846ca68bu;
  x ^= x >> 16;
  return x;
}

__global__ void sample_hist_hi_kernel(const bf16* __restrict__ logits, int V,
                                      unsigned* __restrict__ scratch) {
  __shared__ unsigned h[256];
  logits += (size_t)blockIdx.y * V;
  scratch += (size_t)blockIdx.y * 520;
  const int tid = threadIdx.x;
  if (tid < 256) h[tid] = 0;
  __syncthreads();
  unsigned umax = 0;
  for (int i = blockIdx.x * blockDim.x + tid; i < V;
       i += gridDim.x * blockDim.x) {
    unsigned u = bf16_sortable(
        reinterpret_cast<const unsigned short*>(logits)[i]);
    atomicAdd(&h[u >> 8], 1u);
    umax = max(umax, u);
  }
  __syncthreads();
  if (tid < 256 && h[tid]) atomicAdd(&scratch[tid], h[tid]);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    umax = max(umax, (unsigned)__shfl_xor((int)umax, off, 64));
  if ((tid & 63) == 0) atomicMax(&scratch[515], umax);
}

// decode a sortable u16 code back to the bf16 float value
DEVINL float sortable_to_float(unsigned u) {
  // inverse of bf16_sortable: positive had 0x8000 OR'd, negative was ~x
  unsigned short bits = (u & 0x8000u) ? (unsigned short)(u & 0x7FFFu)
                                      : (unsigned short)(~u & 0xFFFFu);
  bf16 v = *reinterpret_cast<bf16*>(&bits);
  return b2f(v);
}

// ---- top-p (nucleus) mass-radix: exp((l-m)/T) histograms over the same
// sortable buckets; threshold = code where top-down cumulative mass
// crosses top_p * total.  Uses float views of the hist slots (zeroed by
// the preceding count-select kernels).
__global__ void sample_mass_hist_hi_kernel(const bf16* __restrict__ logits,
                                           int V,
                                           unsigned* __restrict__ scratch,
                                           float inv_temp) {
  __shared__ float h[256];
  logits += (size_t)blockIdx.y * V;
  scratch += (size_t)blockIdx.y * 520;
  float* fh = reinterpret_cast<float*>(scratch);
  const int tid = threadIdx.x;
  if (tid < 256) h[tid] = 0.f;
  __syncthreads();
  const float m = sortable_to_float(scratch[515]) * inv_temp;
  // restrict the nucleus mass to codes kept by the preceding top-k pass
  // (scratch[514]): torch composes top-p ON the top-k-masked
  // distribution (models/sampling.py), so the kept sets match
  const unsigned kt = scratch[514];
  for (int i = blockIdx.x * blockDim.x + tid; i < V;
       i += gridDim.x * blockDim.x) {
    unsigned short bits = reinterpret_cast<const unsigned short*>(logits)[i];
    unsigned u = bf16_sortable(bits);
    if (u < kt) continue;
    float l = b2f(*reinterpret_cast<bf16*>(&bits)) * inv_temp;
    atomicAdd(&h[u >> 8], __expf(l - m));
  }
  __syncthreads();
  if (tid < 256 && h[tid] != 0.f) atomicAdd(&fh[tid], h[tid]);
}

__global__ void sample_select_hi_mass_kernel(unsigned* __restrict__ scratch,
                                             float top_p) {
  scratch += (size_t)blockIdx.y * 520;
  float* fh = reinterpret_cast<float*>(scratch);
  if (threadIdx.x != 0) return;
  float total = 0.f;
  for (int b = 0; b < 256; ++b) total += fh[b];
  const float target = top_p * total;
  float cum = 0.f;
  int b = 255;
  for (; b > 0; --b) {
    if (cum + fh[b] >= target) break;
    cum += fh[b];
  }
  scratch[512] = (unsigned)b;  // bucket for the lo pass
  // stash remaining target as float bits in [513]
  float rem = target - cum;
  scratch[513] = __float_as_uint(rem);
}

__global__ void sample_mass_hist_lo_kernel(const bf16* __restrict__ logits,
                                           int V,
                                           unsigned* __restrict__ scratch,
                                           float inv_temp) {
  __shared__ float h[256];
  logits += (size_t)blockIdx.y * V;
  scratch += (size_t)blockIdx.y * 520;
  float* fh = reinterpret_cast<float*>(scratch);
  const int tid = threadIdx.x;
  if (tid < 256) h[tid] = 0.f;
  __syncthreads();
  const unsigned bucket = scratch[512];
  const float m = sortable_to_float(scratch[515]) * inv_temp;
  const unsigned kt = scratch[514];  // top-k threshold (see hi pass)
  for (int i = blockIdx.x * blockDim.x + tid; i < V;
       i += gridDim.x * blockDim.x) {
    unsigned short bits = reinterpret_cast<const unsigned short*>(logits)[i];
    unsigned u = bf16_sortable(bits);
    if (u < kt) continue;
    if ((u >> 8) != bucket) continue;
    float l = b2f(*reinterpret_cast<bf16*>(&bits)) * inv_temp;
    atomicAdd(&h[u & 255], __expf(l - m));
  }
  __syncthreads();
  if (tid < 256 && h[tid] != 0.f) atomicAdd(&fh[256 + tid], h[tid]);
}

__global__ void sample_select_lo_mass_kernel(unsigned* __restrict__ scratch) {
  scratch += (size_t)blockIdx.y * 520;
  float* fh = reinterpret_cast<float*>(scratch);
  if (threadIdx.x != 0) return;
  const float rem = __uint_as_float(scratch[513]);
  const unsigned bucket = scratch[512];
  float cum = 0.f;
  int b = 255;
  for (; b > 0; --b) {
    if (cum + fh[256 + b] >= rem) break;
    cum += fh[256 + b];
  }
  const unsigned tp = (bucket << 8) | (unsigned)b;
  // final threshold = max(count-radix top-k threshold, nucleus threshold)
  if (tp > scratch[514]) scratch[514] = tp;
}

__global__ void sample_select_hi_kernel(unsigned* __restrict__ scratch,
                                        int top_k) {
  // single wave: serial scan from the top bucket down (256 iterations of
  // LDS-free register work — trivial)
  scratch += (size_t)blockIdx.y * 520;
  if (threadIdx.x == 0) {
    unsigned cum = 0;
    int b = 255;
    for (; b >= 0; --b) {
      unsigned c = scratch[b];
      if (cum + c >= (unsigned)top_k) break;
      cum += c;
    }
    if (b < 0) b = 0;
    scratch[512] = (unsigned)b;
    scratch[513] = cum;
  }
}

__global__ void sample_hist_lo_kernel(const bf16* __restrict__ logits, int V,
                                      unsigned* __restrict__ scratch) {
  __shared__ unsigned h[256];
  logits += (size_t)blockIdx.y * V;
  scratch += (size_t)blockIdx.y * 520;
  const int tid = threadIdx.x;
  if (tid < 256) h[tid] = 0;
  __syncthreads();
  const unsigned bucket = scratch[512];
  for (int i = blockIdx.x * blockDim.x + tid; i < V;
       i += gridDim.x * blockDim.x) {
    unsigned u = bf16_sortable(
        reinterpret_cast<const unsigned short*>(logits)[i]);
    if ((u >> 8) == bucket) atomicAdd(&h[u & 255], 1u);
  }
  __syncthreads();
  if (tid < 256 && h[tid]) atomicAdd(&scratch[256 + tid], h[tid]);
}

__global__ void sample_select_lo_kernel(unsigned* __restrict__ scratch,
                                        int top_k) {
  scratch += (size_t)blockIdx.y * 520;
  if (threadIdx.x == 0) {
    unsigned cum = scratch[513];
    const unsigned bucket = scratch[512];
    int b = 255;
    for (; b >= 0; --b) {
      unsigned c = scratch[256 + b];
      if (cum + c >= (unsigned)top_k) break;
      cum += c;
    }
    if (b < 0) b = 0;
    scratch[514] = (bucket << 8) | (unsigned)b;  // threshold: keep u >= t
  }
  __syncthreads();
  // zero the count hists so the (optional) mass phase can reuse the slots
  for (int i = threadIdx.x; i < 512; i += blockDim.x) scratch[i] = 0;
}

__global__ void sample_gumbel_argmax_kernel(
    const bf16* __restrict__ logits, int V, unsigned* __restrict__ scratch,
    float inv_temp, int use_threshold, int noise, unsigned seed,
    const int* __restrict__ pos_p, const int* __restrict__ slot_p,
    int pos_bias) {
  logits += (size_t)blockIdx.y * V;
  scratch += (size_t)blockIdx.y * 520;
  const unsigned t = use_threshold ? scratch[514] : 0u;
  // counter-based RNG keyed by (seed, slot, position of the DRAWN token):
  // reproducible and independent of the scheduling order across samples.
  // pos_bias aligns graphs that read pos before vs after the advance
  // (standalone embed-first step vs pipeline tail-first step).
  const unsigned pos_v =
      pos_p ? (unsigned)(pos_p[blockIdx.y] + pos_bias) : 0u;
  const unsigned slot_v = slot_p ? (unsigned)slot_p[blockIdx.y] : blockIdx.y;
  const unsigned salt = seed ^ (pos_v * 0x9E3779B9u) ^
                        (slot_v * 0x85EBCA6Bu);
  float best = -1e38f;
  int best_i = 0;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < V;
       i += gridDim.x * blockDim.x) {
    unsigned short bits = reinterpret_cast<const unsigned short*>(logits)[i];
    unsigned u = bf16_sortable(bits);
    if (u < t) continue;
    float s = __bfloat162float(*reinterpret_cast<bf16*>(&bits)) * inv_temp;
    if (noise) {
      unsigned h = hash_u32(hash_u32((unsigned)i ^ salt) + salt);
      float uu = (h >> 8) * (1.f / 16777216.f) + 1e-12f;
      s += -logf(-logf(uu));
    }
    if (s > best) {
      best = s;
      best_i = i;
    }
  }
  // wave reduce argmax
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ob = __shfl_xor(best, off, 64);
    int oi = __shfl_xor(best_i, off, 64);
    if (ob > best || (ob == best && oi < best_i)) {
      best = ob;
      best_i = oi;
    }
  }
  if ((threadIdx.x & 63) == 0) {
    unsigned sb = __float_as_uint(best);
    sb = (sb & 0x80000000u) ? ~sb : (sb | 0x80000000u);
    unsigned long long packed =
        ((unsigned long long)sb << 32) | (unsigned)best_i;
    atomicMax(reinterpret_cast<unsigned long long*>(&scratch[516]), packed);
  }
}

// Besides unpacking the winning token and self-cleaning the scratch, this
// kernel optionally performs the decode step's bookkeeping in the same
// launch (single-sample mode): token_table[slot] = token, and
// pos_table[slot] += adv_pos — replacing two torch index kernels in the
// captured step graph.
__global__ void sample_unpack_kernel(unsigned* __restrict__ scratch,
                                     int* __restrict__ out,
                                     int* __restrict__ token_table,
                                     int* __restrict__ pos_table,
                                     const int* __restrict__ slot,
                                     int adv_pos) {
  scratch += (size_t)blockIdx.y * 520;
  out += blockIdx.y;
  if (threadIdx.x == 0) {
    unsigned long long packed =
        *reinterpret_cast<const unsigned long long*>(&scratch[516]);
    const int tok = (int)(unsigned)(packed & 0xFFFFFFFFull);
    out[0] = tok;
    if (slot) {
      const int sl = slot[0];
      if (token_table) token_table[sl] = tok;
      if (pos_table && adv_pos) pos_table[sl] += 1;
    }
  }
  __syncthreads();  // t0's read precedes the clean
  // self-clean the scratch for the next call (single block, runs last)
  for (int i = threadIdx.x; i < 518; i += blockDim.x) scratch[i] = 0;
}

// Step staging in one tiny launch: pos_out = pos_table[slot],
// token_out = token_table[slot], optional pos_table[slot] += 1
// (replaces a copy + two index_selects / an index_add in the graphs).
__global__ void stage_slot_kernel(int* __restrict__ pos_out,
                                  int* __restrict__ token_out,
                                  const int* __restrict__ pos_table,
                                  const int* __restrict__ token_table,
                                  int* __restrict__ pos_table_mut,
                                  const int* __restrict__ slot,
                                  int adv_pos) {
  if (threadIdx.x != 0) return;
  const int sl = slot[0];
  if (pos_out) pos_out[0] = pos_table[sl];
  if (token_out) token_out[0] = token_table[sl];
  if (pos_table_mut && adv_pos) pos_table_mut[sl] += 1;
}

// ---------------------------------------------------------------------------
// Host-side launchers (C ABI used by the torch bindings)
// ---------------------------------------------------------------------------

void launch_sample(void* out_token, const void* logits, int V, void* scratch,
                   float temperature, int top_k, float top_p, int noise_on,
                   unsigned seed, const int* pos, const int* slot,
                   int n_batch, int* token_table, int* pos_table,
                   const int* adv_slot, int adv_pos, int pos_bias,
                   hipStream_t stream) {
  unsigned* sc = (unsigned*)scratch;
  const int B = n_batch > 0 ? n_batch : 1;
  const int blocks = B > 1 ? 32 : 128;
  float inv_t = temperature > 0.f ? 1.f / temperature : 1.f;
  const int use_k = (top_k > 0 && top_k < V) ? 1 : 0;
  const int use_p = (top_p > 0.f && top_p < 1.f) ? 1 : 0;
  if (use_k || use_p) {
    // count radix (also records the global max for the mass phase);
    // with only top-p the count pass still runs (top_k=V keeps everything)
    const int k_eff = use_k ? top_k : V;
    hipLaunchKernelGGL(sample_hist_hi_kernel, dim3(blocks, B), dim3(256), 0,
                       stream, (const bf16*)logits, V, sc);
    hipLaunchKernelGGL(sample_select_hi_kernel, dim3(1, B), dim3(64), 0,
                       stream, sc, k_eff);
    hipLaunchKernelGGL(sample_hist_lo_kernel, dim3(blocks, B), dim3(256), 0,
                       stream, (const bf16*)logits, V, sc);
    hipLaunchKernelGGL(sample_select_lo_kernel, dim3(1, B), dim3(256), 0,
                       stream, sc, k_eff);
  }
  if (use_p) {
    hipLaunchKernelGGL(sample_mass_hist_hi_kernel, dim3(blocks, B),
                       dim3(256), 0, stream, (const bf16*)logits, V, sc,
                       inv_t);
    hipLaunchKernelGGL(sample_select_hi_mass_kernel, dim3(1, B), dim3(64), 0,
                       stream, sc, top_p);
    hipLaunchKernelGGL(sample_mass_hist_lo_kernel, dim3(blocks, B),
                       dim3(256), 0, stream, (const bf16*)logits, V, sc,
                       inv_t);
    hipLaunchKernelGGL(sample_select_lo_mass_kernel, dim3(1, B), dim3(64), 0,
                       stream, sc);
  }
  hipLaunchKernelGGL(sample_gumbel_argmax_kernel, dim3(blocks, B), dim3(256),
                     0, stream, (const bf16*)logits, V, sc, inv_t,
                     use_k || use_p, noise_on, seed, pos, slot, pos_bias);
  hipLaunchKernelGGL(sample_unpack_kernel, dim3(1, B), dim3(64), 0, stream,
                     sc, (int*)out_token, token_table, pos_table, adv_slot,
                     adv_pos);
}

// ---- envelope routing: slot <- hdr[0] on device (pipelined serve) ------
// kind = hdr[2]: 0 data, 1 stop, 2 flush.  Non-data envelopes route to the
// dummy KV slot (scratch compute, nothing real is touched) and reset its
// position so dummy KV writes stay in-bounds.
__global__ void route_env_kernel(const int* __restrict__ hdr,
                                 int* __restrict__ slot_out,
                                 int* __restrict__ pos_table,
                                 int dummy_slot) {
  if (threadIdx.x == 0) {
    int s = hdr[0];
    if (hdr[2] != 0 || s < 0 || s >= dummy_slot) {
      s = dummy_slot;
      pos_table[dummy_slot] = 0;
    }
    slot_out[0] = s;
  }
}

void launch_route_env(const int* hdr, int* slot_out, int* pos_table,
                      int dummy_slot, hipStream_t stream) {
  hipLaunchKernelGGL(route_env_kernel, dim3(1), dim3(64), 0, stream, hdr,
                     slot_out, pos_table, dummy_slot);
}

void launch_stage_slot(int* pos_out, int* token_out, const int* pos_table,
                       const int* token_table, int* pos_table_mut,
                       const int* slot, int adv_pos, hipStream_t stream) {
  hipLaunchKernelGGL(stage_slot_kernel, dim3(1), dim3(64), 0, stream,
                     pos_out, token_out, pos_table, token_table,
                     pos_table_mut, slot, adv_pos);
}

// fused attention+proj launcher.  Returns 0 on success, -1 when the
// geometry has no instantiation (caller falls back to the split path).
template <int QPK, int HS>
static void attn_proj_dispatch2(void* out, const void* qkv, void* kpool,
                                void* vpool, const float* cos_t,
                                const float* sin_t, int rope_ne,
                                const int* pos, const int* slot, int layer,
                                int n_layers_pool, int n_kv_heads,
                                int max_seq, float scale, const void* W,
                                const void* bias, const void* res,
                                void* gran, int M, hipStream_t stream) {
  const int K = n_kv_heads * QPK * HS;
  // rows per block group: fit every block resident at once (2 blocks/CU
  // of 4 waves = the 8-wave/CU limit) AND the W tile in <=79 KB of LDS;
  // grid-stride covers any remainder
  // MDI_ATTN_PROJ_PB caps the number of proj (staging) blocks: fewer
  // co-resident stagers leave the attention blocks' CUs unloaded (the
  // "thin the loader while latency-sensitive work runs" principle,
  // block-granular) at the cost of a longer-but-overlapped stage
  static int pb_cap = -1;
  if (pb_cap < 0) {
    const char* e = getenv("MDI_ATTN_PROJ_PB");
    pb_cap = e ? atoi(e) : 0;
    if (pb_cap <= 0) pb_cap = 2 * 256;
  }
  const int cap_blocks =
      (pb_cap < 2 * 256 ? pb_cap : 2 * 256) - n_kv_heads;
  int BR = (M + cap_blocks - 1) / cap_blocks;
  const int br_lds = (79 * 1024 - 16) / (K * 2);
  if (br_lds < 1) return;  // K too large for the LDS tile (no instantiation)
  if (BR > br_lds) BR = br_lds;
  if (BR < 1) BR = 1;
  int PB = (M + BR - 1) / BR;
  if (PB > cap_blocks) PB = cap_blocks;
  size_t smem_proj = (size_t)BR * K * 2 + 16;  // +verdict word
  size_t smem_attn = (size_t)((17 * HS * 2 + 15) & ~15) +
                     (size_t)ATTN_WAVES * QPK * HS * 4 +
                     (size_t)ATTN_WAVES * QPK * 2 * 4;
  size_t smem = smem_proj > smem_attn ? smem_proj : smem_attn;
  static int thr = -1;
  if (thr < 0) {
    const char* e = getenv("MDI_ATTN_PROJ_SLEEP");
    thr = e ? atoi(e) : 0;
    if (thr != 0 && thr != 4 && thr != 8 && thr != 16) thr = 8;
  }
  static bool attr_set = false;
  if (!attr_set) {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&attn_proj_kernel<QPK, HS, 0>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&attn_proj_kernel<QPK, HS, 4>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&attn_proj_kernel<QPK, HS, 8>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&attn_proj_kernel<QPK, HS, 16>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    attr_set = true;
  }
#define AP_LAUNCH(T)                                                        \
  hipLaunchKernelGGL((attn_proj_kernel<QPK, HS, T>),                        \
                     dim3(n_kv_heads + PB), dim3(256), smem, stream,        \
                     (bf16*)out, (const bf16*)qkv, (bf16*)kpool,            \
                     (bf16*)vpool, cos_t, sin_t, rope_ne, pos, slot,        \
                     layer, n_layers_pool, n_kv_heads, max_seq, scale,      \
                     (const bf16*)W, (const bf16*)bias, (const bf16*)res,   \
                     (unsigned int*)gran, M, BR)
  if (thr == 0) AP_LAUNCH(0);
  else if (thr == 4) AP_LAUNCH(4);
  else if (thr == 8) AP_LAUNCH(8);
  else AP_LAUNCH(16);
#undef AP_LAUNCH
}

template <int QPK>
static int attn_proj_dispatch1(int head_size, void* out, const void* qkv,
                               void* kpool, void* vpool, const float* cos_t,
                               const float* sin_t, int rope_ne,
                               const int* pos, const int* slot, int layer,
                               int n_layers_pool, int n_kv_heads,
                               int max_seq, float scale, const void* W,
                               const void* bias, const void* res, void* gran,
                               int M, hipStream_t stream) {
#define CASE_HS_P(H)                                                        \
  if (head_size == H) {                                                     \
    if constexpr (QPK * H >= 64) {                                          \
      attn_proj_dispatch2<QPK, H>(out, qkv, kpool, vpool, cos_t, sin_t,     \
                                  rope_ne, pos, slot, layer, n_layers_pool, \
                                  n_kv_heads, max_seq, scale, W, bias, res, \
                                  gran, M, stream);                         \
      return 0;                                                             \
    }                                                                       \
  }
  CASE_HS_P(64)
  CASE_HS_P(128)
  CASE_HS_P(256)
#undef CASE_HS_P
  return -1;
}

int launch_attn_proj(void* out, const void* qkv, void* kpool, void* vpool,
                     const float* cos_t, const float* sin_t, int rope_ne,
                     const int* pos, const int* slot, int layer,
                     int n_layers_pool, int n_kv_heads, int max_seq,
                     int head_size, int qpk, float scale, const void* W,
                     const void* bias, const void* res, void* gran, int M,
                     hipStream_t stream) {
  if (max_seq > ATTN_BLOCK_MAX_SEQ) return -1;
  switch (qpk) {
    case 1:
      return attn_proj_dispatch1<1>(head_size, out, qkv, kpool, vpool,
                                    cos_t, sin_t, rope_ne, pos, slot, layer,
                                    n_layers_pool, n_kv_heads, max_seq,
                                    scale, W, bias, res, gran, M, stream);
    case 2:
      return attn_proj_dispatch1<2>(head_size, out, qkv, kpool, vpool,
                                    cos_t, sin_t, rope_ne, pos, slot, layer,
                                    n_layers_pool, n_kv_heads, max_seq,
                                    scale, W, bias, res, gran, M, stream);
    case 4:
      return attn_proj_dispatch1<4>(head_size, out, qkv, kpool, vpool,
                                    cos_t, sin_t, rope_ne, pos, slot, layer,
                                    n_layers_pool, n_kv_heads, max_seq,
                                    scale, W, bias, res, gran, M, stream);
    case 8:
      return attn_proj_dispatch1<8>(head_size, out, qkv, kpool, vpool,
                                    cos_t, sin_t, rope_ne, pos, slot, layer,
                                    n_layers_pool, n_kv_heads, max_seq,
                                    scale, W, bias, res, gran, M, stream);
    case 16:
      return attn_proj_dispatch1<16>(head_size, out, qkv, kpool, vpool,
                                     cos_t, sin_t, rope_ne, pos, slot,
                                     layer, n_layers_pool, n_kv_heads,
                                     max_seq, scale, W, bias, res, gran, M,
                                     stream);
  }
  return -1;
}

// grouped M-tile GEMM launcher; returns -1 when the shape has no
// instantiation (caller falls back to hipBLASLt)
int launch_mtile_gemm(void* Y, const void* W, const void* X,
                      const void* bias, const void* res, int Bsz, int M,
                      int K, hipStream_t stream) {
  if (M % 16 != 0) return -1;
  const int grid = M / 16;
#define MT_CASE(BB, KCC)                                                    \
  if (Bsz == BB) {                                                          \
    if (K % (2 * KCC) != 0 || K < 2 * KCC) return -1;                             \
    static bool attr_set_##BB = false;                                      \
    if (!attr_set_##BB) {                                                   \
      (void)hipFuncSetAttribute(                                            \
          reinterpret_cast<const void*>(&mtile_gemm_kernel<BB>),            \
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);          \
      attr_set_##BB = true;                                                 \
    }                                                                       \
    hipLaunchKernelGGL((mtile_gemm_kernel<BB>), dim3(grid), dim3(256),      \
                       (size_t)2 * BB * KCC * 2, stream, (bf16*)Y,          \
                       (const bf16*)W, (const bf16*)X, (const bf16*)bias,   \
                       (const bf16*)res, M, K);                             \
    return 0;                                                               \
  }
  MT_CASE(16, 512)
  MT_CASE(32, 512)
  MT_CASE(64, 256)
  MT_CASE(128, 128)
#undef MT_CASE
  return -1;
}

static inline int gemv_grid(int M, int rows_per_block) {
  int blocks = (M + rows_per_block - 1) / rows_per_block;
  // optional cap: fewer, longer-lived waves (each loops rows, letting the
  // next row's loads pipeline over the current row's tail) — sweep knob
  static int cap = -1;
  if (cap < 0) {
    const char* e = getenv("MDI_GEMV_GRID_CAP");
    cap = e ? atoi(e) : 4096;
    if (cap <= 0) cap = 4096;
  }
  return blocks < cap ? blocks : cap;
}

void launch_rmsnorm(void* out, const void* x, const void* w, int n, float eps,
                    int n_batch, hipStream_t stream) {
  hipLaunchKernelGGL(rmsnorm_kernel, dim3(1, n_batch > 0 ? n_batch : 1),
                     dim3(256), 0, stream, (bf16*)out, (const bf16*)x,
                     (const bf16*)w, n, eps);
}

void launch_layernorm(void* out, const void* x, const void* w, const void* b,
                      int n, float eps, hipStream_t stream) {
  hipLaunchKernelGGL(layernorm_kernel, dim3(1), dim3(256), 0, stream,
                     (bf16*)out, (const bf16*)x, (const bf16*)w,
                     (const bf16*)b, n, eps);
}

void launch_gemv(void* out, const void* W, const void* x, const void* bias,
                 const void* res, const void* norm_w, const void* norm_b,
                 float eps, int M, int K, int epilogue, int norm_kind,
                 int rows, const int* eidx, long long estride,
                 hipStream_t stream) {
  const int smem = K * sizeof(bf16);
  if (rows == 0) rows = M >= 32768 ? 4 : (M > 8192 ? 2 : 1);
  dim3 grid(gemv_grid(M, 4 * rows)), block(256);
  // register-staged fused-RMS form for the qkv shape, grid shrunk so the
  // x*nw staging prologue amortizes over ~3 rows per wave (see kernel)
  static int pre_on = -1;
  if (pre_on < 0) {
    const char* e = getenv("MDI_GEMV_PRE");
    pre_on = e ? atoi(e) : 0;
  }
  if (pre_on && epilogue == 0 && norm_kind == 1 && rows == 1 &&
      eidx == nullptr && res == nullptr && K >= 512 && K % 512 == 0 &&
      K <= 8192 && M % 4 == 0) {
    dim3 pgrid(gemv_grid(M, 4 * 3));
    if (K <= 4096)
      hipLaunchKernelGGL((gemv_direct_pre_kernel<8>), pgrid, block, 0,
                         stream, (bf16*)out, (const bf16*)W,
                         (const bf16*)x, (const bf16*)bias,
                         (const bf16*)norm_w, eps, M, K);
    else
      hipLaunchKernelGGL((gemv_direct_pre_kernel<16>), pgrid, block, 0,
                         stream, (bf16*)out, (const bf16*)W,
                         (const bf16*)x, (const bf16*)bias,
                         (const bf16*)norm_w, eps, M, K);
    return;
  }
  // NORM 0/1: the direct-x kernel (no staging barrier); 2: staged LDS form
#define GEMV_CASE1(E, N, R)                                                 \
  do {                                                                      \
    if (N == 0 && K <= 6144 && R == 1)                                      \
      hipLaunchKernelGGL((gemv_direct_kernel<E, N == 2 ? 0 : N, R, 4>),     \
                         grid, block, 0, stream, (bf16*)out,                \
                         (const bf16*)W, (const bf16*)x, (const bf16*)bias, \
                         (const bf16*)res, (const bf16*)norm_w, eps, M, K,  \
                         eidx, estride);                                    \
    else if (N != 2)                                                        \
      hipLaunchKernelGGL((gemv_direct_kernel<E, N == 2 ? 0 : N, R, 2>),     \
                         grid, block, 0, stream, (bf16*)out,                \
                         (const bf16*)W, (const bf16*)x, (const bf16*)bias, \
                         (const bf16*)res, (const bf16*)norm_w, eps, M, K,  \
                         eidx, estride);                                    \
    else                                                                    \
      hipLaunchKernelGGL((gemv_kernel<E, N, R>), grid, block, smem, stream, \
                         (bf16*)out, (const bf16*)W, (const bf16*)x,        \
                         (const bf16*)bias, (const bf16*)res,               \
                         (const bf16*)norm_w, (const bf16*)norm_b, eps, M,  \
                         K);                                                \
  } while (0)
#define GEMV_CASE(E, N)                                                     \
  do {                                                                      \
    if (rows == 4) GEMV_CASE1(E, N, 4);                                     \
    else if (rows == 2) GEMV_CASE1(E, N, 2);                                \
    else GEMV_CASE1(E, N, 1);                                               \
  } while (0)
  switch (epilogue * 4 + norm_kind) {
    case 0: GEMV_CASE(0, 0); break;
    case 1: GEMV_CASE(0, 1); break;
    case 2: GEMV_CASE(0, 2); break;
    case 4: GEMV_CASE(1, 0); break;
    case 5: GEMV_CASE(1, 1); break;
    case 6: GEMV_CASE(1, 2); break;
    case 8: GEMV_CASE(2, 0); break;
    case 9: GEMV_CASE(2, 1); break;
    case 10: GEMV_CASE(2, 2); break;
    case 12: GEMV_CASE(3, 0); break;
    case 13: GEMV_CASE(3, 1); break;
    case 14: GEMV_CASE(3, 2); break;
    default: GEMV_CASE(0, 0);
  }
#undef GEMV_CASE
#undef GEMV_CASE1
}

void launch_gemv_fp8(void* out, const void* W, const float* wscale,
                     const void* x, const void* bias, const void* res,
                     const void* norm_w, const void* norm_b, float eps,
                     int M, int K, int epilogue, int norm_kind, int rows,
                     hipStream_t stream) {
  const int smem = K * sizeof(bf16);
  if (rows == 0) rows = M >= 32768 ? 4 : (M > 8192 ? 2 : 1);
  dim3 grid(gemv_grid(M, 4 * rows)), block(256);
#define GEMV8_CASE1(E, N, R)                                                \
  hipLaunchKernelGGL((gemv_fp8_kernel<E, N, R>), grid, block, smem, stream, \
                     (bf16*)out, (const unsigned char*)W, wscale,           \
                     (const bf16*)x, (const bf16*)bias, (const bf16*)res,   \
                     (const bf16*)norm_w, (const bf16*)norm_b, eps, M, K)
#define GEMV8_CASE(E, N)                                                    \
  do {                                                                      \
    if (rows == 4) GEMV8_CASE1(E, N, 4);                                    \
    else if (rows == 2) GEMV8_CASE1(E, N, 2);                               \
    else GEMV8_CASE1(E, N, 1);                                              \
  } while (0)
  switch (epilogue * 4 + norm_kind) {
    case 0: GEMV8_CASE(0, 0); break;
    case 1: GEMV8_CASE(0, 1); break;
    case 2: GEMV8_CASE(0, 2); break;
    case 4: GEMV8_CASE(1, 0); break;
    case 5: GEMV8_CASE(1, 1); break;
    case 6: GEMV8_CASE(1, 2); break;
    case 8: GEMV8_CASE(2, 0); break;
    case 9: GEMV8_CASE(2, 1); break;
    case 10: GEMV8_CASE(2, 2); break;
    case 12: GEMV8_CASE(3, 0); break;
    case 13: GEMV8_CASE(3, 1); break;
    case 14: GEMV8_CASE(3, 2); break;
    default: GEMV8_CASE(0, 0);
  }
#undef GEMV8_CASE
#undef GEMV8_CASE1
}

void launch_gemv_swiglu_fp8(void* out, const void* Wg, const float* gscale,
                            const void* Wu, const float* uscale,
                            const void* x, const void* norm_w,
                            const void* norm_b, float eps, int M, int K,
                            int gelu_gate, int norm_kind,
                            hipStream_t stream) {
  const int smem = K * sizeof(bf16);
  dim3 grid(gemv_grid(M, 4)), block(256);
#define SW8_CASE(N)                                                         \
  hipLaunchKernelGGL((gemv_swiglu_fp8_kernel<N>), grid, block, smem,        \
                     stream, (bf16*)out, (const unsigned char*)Wg, gscale,  \
                     (const unsigned char*)Wu, uscale, (const bf16*)x,      \
                     (const bf16*)norm_w, (const bf16*)norm_b, eps, M, K,   \
                     gelu_gate)
  switch (norm_kind) {
    case 1: SW8_CASE(1); break;
    case 2: SW8_CASE(2); break;
    default: SW8_CASE(0);
  }
#undef SW8_CASE
}

void launch_gemv_swiglu(void* out, const void* Wg, const void* Wu,
                        const void* x, const void* norm_w, const void* norm_b,
                        float eps, int M, int K, int gelu_gate, int norm_kind,
                        const int* eidx, long long estride,
                        const float* escale, hipStream_t stream) {
  const int smem = K * sizeof(bf16);
  dim3 grid(gemv_grid(M, 4)), block(256);
#define SW_CASE(N)                                                          \
  hipLaunchKernelGGL((gemv_swiglu_kernel<N>), grid, block, smem, stream,    \
                     (bf16*)out, (const bf16*)Wg, (const bf16*)Wu,          \
                     (const bf16*)x, (const bf16*)norm_w,                   \
                     (const bf16*)norm_b, eps, M, K, gelu_gate, eidx,       \
                     estride, escale)
  switch (norm_kind) {
    case 1: SW_CASE(1); break;
    case 2: SW_CASE(2); break;
    default: SW_CASE(0);
  }
#undef SW_CASE
}

void launch_swiglu_mul(void* out, const void* g, const void* u,
                       long long n, int gelu_gate, hipStream_t stream) {
  long long blocks = (n / 8 + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(swiglu_mul_kernel, dim3((unsigned)blocks), dim3(256),
                     0, stream, (bf16*)out, (const bf16*)g, (const bf16*)u,
                     n, gelu_gate);
}

void launch_moe_gate_topk(int* eidx, float* escale, const void* logits,
                          int n_e, int k, hipStream_t stream) {
  hipLaunchKernelGGL(moe_gate_topk_kernel, dim3(1), dim3(64), 0, stream,
                     eidx, escale, (const bf16*)logits, n_e, k);
}

void launch_embed(void* out, const void* wte, const int* token, int n_embd,
                  float scale, hipStream_t stream) {
  int blocks = (n_embd / 8 + 255) / 256;
  hipLaunchKernelGGL(embed_kernel, dim3(blocks > 0 ? blocks : 1), dim3(256),
                     0, stream, (bf16*)out, (const bf16*)wte, token, n_embd,
                     scale);
}

void launch_rope_kv_append(void* qkv, void* kpool, void* vpool,
                           const float* cos_t, const float* sin_t,
                           const int* pos, const int* slot, int layer,
                           int n_layers_pool, int n_kv_heads, int max_seq,
                           int head_size, int rope_n_elem, int qpk,
                           hipStream_t stream) {
  hipLaunchKernelGGL(rope_kv_append_kernel, dim3(n_kv_heads), dim3(256), 0,
                     stream, (bf16*)qkv, (bf16*)kpool, (bf16*)vpool, cos_t,
                     sin_t, pos, slot, layer, n_layers_pool, n_kv_heads,
                     max_seq, head_size, rope_n_elem, qpk);
}

template <int QPK, int HS>
static void attn_dispatch2(void* out, float* part_o, float* part_ml,
                           const void* qkv, void* kpool, void* vpool,
                           float* kscale, float* vscale,
                           const float* cos_t, const float* sin_t,
                           int rope_ne, const int* pos, const int* slot,
                           int layer, int n_layers_pool, int n_kv_heads,
                           int max_seq, int n_chunks, float scale,
                           int n_batch, int blocks, int force_split,
                           hipStream_t stream) {
  const int kv8 = kscale != nullptr;
  // short/medium contexts: one block per (batch, kv_head), no global
  // partials, no combine kernel — the whole attention step is ONE launch.
  // MDI_ATTN_FORCE_SPLITS=1 forces the split-S + combine path instead
  // (A/B knob: the round-2 split-S got block-cooperative staging and a
  // parallel combine, so its short-S cost may have changed)
  static int force_splits = -1;
  if (force_splits < 0) {
    const char* e = getenv("MDI_ATTN_FORCE_SPLITS");
    force_splits = (e && e[0] == '1') ? 1 : 0;
  }
  if (max_seq <= ATTN_BLOCK_MAX_SEQ && !force_splits && !force_split) {
    const int nb = (n_batch > 0 ? n_batch : 1) * n_kv_heads;
#define ATTN_BLK(BATCHV, KV8V, NBATCH)                                      \
    hipLaunchKernelGGL((attn_decode_block_kernel<QPK, HS, BATCHV, KV8V>),   \
                       dim3(nb), dim3(256), 0, stream, (bf16*)out,          \
                       (const bf16*)qkv, kpool, vpool, kscale, vscale,      \
                       cos_t, sin_t, rope_ne, pos, slot, layer,             \
                       n_layers_pool, n_kv_heads, max_seq, scale, NBATCH)
    if (n_batch > 0) {
      if (kv8) ATTN_BLK(1, 1, n_batch);
      else ATTN_BLK(1, 0, n_batch);
    } else {
      if (kv8) ATTN_BLK(0, 1, 1);
      else ATTN_BLK(0, 0, 1);
    }
#undef ATTN_BLK
    return;
  }
#define ATTN_SPL(BATCHV, KV8V, NBATCH)                                      \
  hipLaunchKernelGGL((attn_decode_kernel<QPK, HS, BATCHV, KV8V>),           \
                     dim3(blocks), dim3(256), 0, stream, part_o, part_ml,   \
                     (const bf16*)qkv, kpool, vpool, kscale, vscale, cos_t, \
                     sin_t, rope_ne, pos, slot, layer, n_layers_pool,       \
                     n_kv_heads, max_seq, n_chunks, scale, NBATCH)
  if (n_batch > 0) {
    if (kv8) ATTN_SPL(1, 1, n_batch);
    else ATTN_SPL(1, 0, n_batch);
  } else {
    if (kv8) ATTN_SPL(0, 1, 1);
    else ATTN_SPL(0, 0, 1);
  }
#undef ATTN_SPL
}

template <int QPK>
static int attn_dispatch1(int head_size, void* out, float* part_o,
                          float* part_ml, const void* qkv, void* kpool,
                          void* vpool, float* kscale, float* vscale,
                          const float* cos_t,
                          const float* sin_t, int rope_ne, const int* pos,
                          const int* slot, int layer, int n_layers_pool,
                          int n_kv_heads, int max_seq, int n_chunks,
                          float scale, int n_batch, int blocks,
                          int force_split, hipStream_t stream) {
#define CASE_HS(H)                                                          \
  if (head_size == H) {                                                     \
    if constexpr (QPK * H >= 64) {                                          \
      attn_dispatch2<QPK, H>(out, part_o, part_ml, qkv, kpool, vpool,       \
                             kscale, vscale, cos_t, sin_t, rope_ne, pos,    \
                             slot, layer, n_layers_pool, n_kv_heads,        \
                             max_seq, n_chunks, scale, n_batch, blocks,     \
                             force_split, stream);                          \
      return 0;                                                             \
    }                                                                       \
  }
  CASE_HS(64)
  CASE_HS(128)
  CASE_HS(256)
#undef CASE_HS
  return -1;  // unsupported geometry: caller falls back to the torch path
}

// returns 0 on success, -1 if (qpk, head_size) has no kernel instantiation
// n_batch == 0 -> single-token mode; > 0 -> batched (pos/slot arrays)
int launch_attn_decode(void* out, float* part_o, float* part_ml,
                       const void* qkv, void* kpool, void* vpool,
                       float* kscale, float* vscale,
                       const float* cos_t, const float* sin_t, int rope_ne,
                       const int* pos, const int* slot, int layer,
                       int n_layers_pool, int n_kv_heads, int max_seq,
                       int head_size, int qpk, int n_chunks, float scale,
                       int n_batch, int force_split, hipStream_t stream) {
  const int n_wg = n_kv_heads * n_chunks * (n_batch > 0 ? n_batch : 1);
  const int blocks = (n_wg + ATTN_WAVES - 1) / ATTN_WAVES;
  int rc = -1;
  switch (qpk) {
    case 1:
      rc = attn_dispatch1<1>(
          head_size, out, part_o, part_ml, qkv, kpool, vpool, kscale,
          vscale, cos_t, sin_t,
          rope_ne, pos, slot, layer, n_layers_pool, n_kv_heads, max_seq,
          n_chunks, scale, n_batch, blocks, force_split, stream);
      break;
    case 2:
      rc = attn_dispatch1<2>(
          head_size, out, part_o, part_ml, qkv, kpool, vpool, kscale,
          vscale, cos_t, sin_t,
          rope_ne, pos, slot, layer, n_layers_pool, n_kv_heads, max_seq,
          n_chunks, scale, n_batch, blocks, force_split, stream);
      break;
    case 4:
      rc = attn_dispatch1<4>(
          head_size, out, part_o, part_ml, qkv, kpool, vpool, kscale,
          vscale, cos_t, sin_t,
          rope_ne, pos, slot, layer, n_layers_pool, n_kv_heads, max_seq,
          n_chunks, scale, n_batch, blocks, force_split, stream);
      break;
    case 8:
      rc = attn_dispatch1<8>(
          head_size, out, part_o, part_ml, qkv, kpool, vpool, kscale,
          vscale, cos_t, sin_t,
          rope_ne, pos, slot, layer, n_layers_pool, n_kv_heads, max_seq,
          n_chunks, scale, n_batch, blocks, force_split, stream);
      break;
    case 16:
      rc = attn_dispatch1<16>(
          head_size, out, part_o, part_ml, qkv, kpool, vpool, kscale,
          vscale, cos_t, sin_t,
          rope_ne, pos, slot, layer, n_layers_pool, n_kv_heads, max_seq,
          n_chunks, scale, n_batch, blocks, force_split, stream);
      break;
    default:
      return -1;
  }
  if (rc != 0) return rc;
  static int force_splits2 = -1;
  if (force_splits2 < 0) {
    const char* e = getenv("MDI_ATTN_FORCE_SPLITS");
    force_splits2 = (e && e[0] == '1') ? 1 : 0;
  }
  if (max_seq <= ATTN_BLOCK_MAX_SEQ && !force_splits2 && !force_split)
    return 0;  // block-local: no combine
  // the combine kernel is batch-agnostic: [B, n_head, chunks, hs] is just
  // B*n_head heads
  const int n_head_eff = n_kv_heads * qpk * (n_batch > 0 ? n_batch : 1);
  const int ds = (head_size + 63) / 64;
  const int cblocks = n_head_eff * ds;  // one block per (head, dim slice)
  hipLaunchKernelGGL(attn_combine_kernel, dim3(cblocks), dim3(256), 0,
                     stream, (bf16*)out, part_o, part_ml, n_chunks, head_size,
                     n_head_eff, pos, n_kv_heads * qpk);
  return 0;
}

void launch_rope_prefill_append(void* qkv, void* kpool, void* vpool,
                                float* kscale, float* vscale,
                                const float* cos_t, const float* sin_t,
                                int pos0, int slot, int layer,
                                int n_layers_pool, int n_kv_heads,
                                int max_seq, int head_size, int rope_ne,
                                int qpk, int T, hipStream_t stream) {
  if (kscale != nullptr)
    hipLaunchKernelGGL((rope_prefill_append_kernel<1>),
                       dim3(n_kv_heads, T), dim3(256), 0, stream,
                       (bf16*)qkv, kpool, vpool, kscale, vscale, cos_t,
                       sin_t, pos0, slot, layer, n_layers_pool, n_kv_heads,
                       max_seq, head_size, rope_ne, qpk);
  else
    hipLaunchKernelGGL((rope_prefill_append_kernel<0>),
                       dim3(n_kv_heads, T), dim3(256), 0, stream,
                       (bf16*)qkv, kpool, vpool, kscale, vscale, cos_t,
                       sin_t, pos0, slot, layer, n_layers_pool, n_kv_heads,
                       max_seq, head_size, rope_ne, qpk);
}

int launch_prefill_attn(void* out, const void* qkv, const void* kpool,
                        const void* vpool, const float* kscale,
                        const float* vscale, int pos0, int slot, int layer,
                        int n_layers_pool, int n_kv_heads, int max_seq,
                        int head_size, int qpk, int T, float scale,
                        hipStream_t stream) {
  const int n_head = n_kv_heads * qpk;
  const int n_qtiles = (T + 15) / 16;
  const int blocks = (n_head * n_qtiles + ATTN_WAVES - 1) / ATTN_WAVES;
  const int kv8 = kscale != nullptr;
  static const bool v1 = [] {
    const char* e = getenv("MDI_PREFILL_V1");
    return e != nullptr && e[0] == '1';
  }();
  // chunk-size A/B knob: 64 (2 blocks/CU at HS=128) vs 32 (4 blocks/CU,
  // twice the barriers) — 64 measured faster, see profiles
  static const int kch = [] {
    const char* e = getenv("MDI_PREFILL_KCH");
    return (e != nullptr && atoi(e) == 32) ? 32 : 64;
  }();
  if (!v1) {
    // v2: workgroup shares K/V staging across the GQA group; P.V by MFMA
    const int hpb = qpk >= 4 ? 4 : qpk;  // q heads per block
    const int qtpb = (hpb == 1) ? 4 : (hpb == 2 ? 2 : 1);  // q tiles / block
    const int n_hgrp = (qpk + hpb - 1) / hpb;
    const int n_qtg = (n_qtiles + qtpb - 1) / qtpb;
    const int blocks2 = n_kv_heads * n_hgrp * n_qtg;
#define PF2_LAUNCH(H, K8, KC)                                               \
  hipLaunchKernelGGL((prefill_attn_mfma_kernel<H, K8, KC>), dim3(blocks2),  \
                     dim3(256), 0, stream, (bf16*)out, (const bf16*)qkv,    \
                     kpool, vpool, kscale, vscale, pos0, slot, layer,       \
                     n_layers_pool, n_kv_heads, max_seq, qpk, T, scale,     \
                     hpb, qtpb, n_hgrp, n_qtg)
#define PF2_CASE(H)                                                         \
  if (head_size == H) {                                                     \
    if (kv8) {                                                              \
      if (kch == 32) PF2_LAUNCH(H, 1, 32); else PF2_LAUNCH(H, 1, 64);       \
    } else {                                                                \
      if (kch == 32) PF2_LAUNCH(H, 0, 32); else PF2_LAUNCH(H, 0, 64);       \
    }                                                                       \
    return 0;                                                               \
  }
    PF2_CASE(64)
    PF2_CASE(128)
    PF2_CASE(256)
#undef PF2_CASE
#undef PF2_LAUNCH
  }
#define PF_CASE(H)                                                          \
  if (head_size == H) {                                                     \
    if (kv8)                                                                \
      hipLaunchKernelGGL((prefill_attn_kernel<H, 1>), dim3(blocks),         \
                         dim3(256), 0, stream, (bf16*)out,                  \
                         (const bf16*)qkv, kpool, vpool, kscale, vscale,    \
                         pos0, slot, layer, n_layers_pool, n_kv_heads,      \
                         max_seq, qpk, T, scale);                           \
    else                                                                    \
      hipLaunchKernelGGL((prefill_attn_kernel<H, 0>), dim3(blocks),         \
                         dim3(256), 0, stream, (bf16*)out,                  \
                         (const bf16*)qkv, kpool, vpool, kscale, vscale,    \
                         pos0, slot, layer, n_layers_pool, n_kv_heads,      \
                         max_seq, qpk, T, scale);                           \
    return 0;                                                               \
  }
  PF_CASE(64)
  PF_CASE(128)
  PF_CASE(256)
#undef PF_CASE
  return -1;
}

void launch_add(void* out, const void* a, const void* b, int n,
                hipStream_t stream) {
  int blocks = (n / 8 + 255) / 256;
  hipLaunchKernelGGL(add_kernel, dim3(blocks > 0 ? blocks : 1), dim3(256), 0,
                     stream, (bf16*)out, (const bf16*)a, (const bf16*)b, n);
}
