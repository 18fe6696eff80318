// Grouped (gather-style) expert MLP kernels for the stacked `switch_mlp`
// layout (/root/reference/shard/server/model/deepseek_v2.py:101-112).
//
// Decode regime: N tokens × top_k experts.  Tokens are sorted by expert
// on the host and split into sub-ranges of ≤ MG_TOK tokens; each block
// owns one (sub-range, output-row tile) and streams the expert's weight
// rows ONCE, applying each row to all MG_TOK staged tokens — expert
// weights are read per *expert sub-range*, not per (token, expert) pair
// (the naive gather re-read weights per pair: 96% of decode GPU time).
//
// x for the sub-range is staged in LDS (MG_TOK * H bf16); per-token dot
// accumulators are statically indexed (guide §5.4 rule 20).  Large-N
// (prefill) goes through per-expert hipBLASLt GEMMs from Python.

#include "hip_common.h"

#define MG_BLOCK 256
#define MG_WAVES (MG_BLOCK / WAVE)
#define MG_TOK 4

// ---------------------------------------------------------------------------
// bf16 experts: fused gate/up + SwiGLU.
// grid = (row_tiles, n_subranges)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(MG_BLOCK) void moe_gateup_grouped_kernel(
    const short* __restrict__ x,        // [N, H]
    const short* __restrict__ gate_w,   // [E, I, H]
    const short* __restrict__ up_w,     // [E, I, H]
    short* __restrict__ h,              // [P, I] (sorted pair order)
    const int* __restrict__ sub_expert, // [S]
    const int* __restrict__ sub_off,    // [S] first sorted-pair index
    const int* __restrict__ sub_cnt,    // [S] tokens in sub-range (<= MG_TOK)
    const int* __restrict__ sorted_tok, // [P]
    int H, int I) {
  const int s = blockIdx.y;
  const int e = sub_expert[s];
  const int p0 = sub_off[s];
  const int cnt = sub_cnt[s];
  if (cnt == 0) return;  // padded slot
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  short* x_lds = reinterpret_cast<short*>(smem_raw);  // [MG_TOK][H]
#pragma unroll
  for (int t = 0; t < MG_TOK; ++t) {
    short4v* dst = reinterpret_cast<short4v*>(x_lds + t * H);
    if (t < cnt) {
      const short4v* src =
          reinterpret_cast<const short4v*>(x + (long)sorted_tok[p0 + t] * H);
      for (int i = tid; i < H / 4; i += MG_BLOCK) dst[i] = src[i];
    } else {
      for (int i = tid; i < H / 4; i += MG_BLOCK) dst[i] = short4v{0, 0, 0, 0};
    }
  }
  __syncthreads();

  const long ebase = (long)e * I * H;
  for (int o = blockIdx.x * MG_WAVES + wid; o < I; o += gridDim.x * MG_WAVES) {
    const short* grow = gate_w + ebase + (long)o * H;
    const short* urow = up_w + ebase + (long)o * H;
    float gdot[MG_TOK] = {};
    float udot[MG_TOK] = {};
    // 16 B/lane weight loads (guide G13); v_dot2c_f32_bf16 does 2
    // products + accumulate per VALU op (4 instr per 8-elem chunk per
    // operand vs 24 cvt+fma before) — the scalar kernels were VALU-
    // limited at ~4 TB/s, this lifts the per-token compute cost.
    for (int d = lane * 8; d < H; d += WAVE * 8) {
      bf16x8_t gv = *reinterpret_cast<const bf16x8_t*>(grow + d);
      bf16x8_t uv = *reinterpret_cast<const bf16x8_t*>(urow + d);
#pragma unroll
      for (int t = 0; t < MG_TOK; ++t) {
        bf16x8_t xv = *reinterpret_cast<const bf16x8_t*>(x_lds + t * H + d);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          bf16x2_t xp = {xv[2 * j], xv[2 * j + 1]};
          gdot[t] = dot2_bf16(bf16x2_t{gv[2 * j], gv[2 * j + 1]}, xp, gdot[t]);
          udot[t] = dot2_bf16(bf16x2_t{uv[2 * j], uv[2 * j + 1]}, xp, udot[t]);
        }
      }
    }
#pragma unroll
    for (int t = 0; t < MG_TOK; ++t) {
      float g = wave_sum(gdot[t]);
      float u = wave_sum(udot[t]);
      if (lane == 0 && t < cnt) {
        float a = g / (1.0f + __expf(-g));  // silu
        h[(long)(p0 + t) * I + o] = (short)__bfloat16_as_ushort(f2bf(a * u));
      }
    }
  }
}

// Down-proj + weighted atomic scatter into fp32 out.
__global__ __launch_bounds__(MG_BLOCK) void moe_down_grouped_kernel(
    const short* __restrict__ h,        // [P, I] sorted pair order
    const short* __restrict__ down_w,   // [E, H, I]
    float* __restrict__ out,            // [N, H] fp32 (pre-zeroed)
    const int* __restrict__ sub_expert, const int* __restrict__ sub_off,
    const int* __restrict__ sub_cnt, const int* __restrict__ sorted_tok,
    const float* __restrict__ sorted_wt,  // [P]
    int I, int H) {
  const int s = blockIdx.y;
  const int e = sub_expert[s];
  const int p0 = sub_off[s];
  const int cnt = sub_cnt[s];
  if (cnt == 0) return;  // padded slot
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  short* h_lds = reinterpret_cast<short*>(smem_raw);  // [MG_TOK][I]
#pragma unroll
  for (int t = 0; t < MG_TOK; ++t) {
    short4v* dst = reinterpret_cast<short4v*>(h_lds + t * I);
    if (t < cnt) {
      const short4v* src =
          reinterpret_cast<const short4v*>(h + (long)(p0 + t) * I);
      for (int i = tid; i < I / 4; i += MG_BLOCK) dst[i] = src[i];
    } else {
      for (int i = tid; i < I / 4; i += MG_BLOCK) dst[i] = short4v{0, 0, 0, 0};
    }
  }
  __syncthreads();

  const long ebase = (long)e * H * I;
  for (int o = blockIdx.x * MG_WAVES + wid; o < H; o += gridDim.x * MG_WAVES) {
    const short* drow = down_w + ebase + (long)o * I;
    float dot[MG_TOK] = {};
    // 16 B/lane weight loads (guide G13); v_dot2c accumulate as in the
    // gate/up kernel.  I % 512 handled by the 8-tail.
    for (int d = lane * 8; d + 7 < I; d += WAVE * 8) {
      bf16x8_t dv = *reinterpret_cast<const bf16x8_t*>(drow + d);
#pragma unroll
      for (int t = 0; t < MG_TOK; ++t) {
        bf16x8_t hv = *reinterpret_cast<const bf16x8_t*>(h_lds + t * I + d);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          dot[t] = dot2_bf16(bf16x2_t{dv[2 * j], dv[2 * j + 1]},
                             bf16x2_t{hv[2 * j], hv[2 * j + 1]}, dot[t]);
      }
    }
#pragma unroll
    for (int t = 0; t < MG_TOK; ++t) {
      float v = wave_sum(dot[t]);
      if (lane == 0 && t < cnt)
        atomicAdd(out + (long)sorted_tok[p0 + t] * H + o, sorted_wt[p0 + t] * v);
    }
  }
}

extern "C" void launch_moe_gateup_grouped(const void* x, const void* gate_w,
                                          const void* up_w, void* h,
                                          const int* sub_expert,
                                          const int* sub_off,
                                          const int* sub_cnt,
                                          const int* sorted_tok, int S, int H,
                                          int I, hipStream_t stream) {
  size_t smem = (size_t)MG_TOK * H * sizeof(short);
  int gx = (I + MG_WAVES - 1) / MG_WAVES;
  if (gx > 64) gx = 64;  // rows loop inside the block: amortize x staging
  moe_gateup_grouped_kernel<<<dim3(gx, S), dim3(MG_BLOCK), smem, stream>>>(
      (const short*)x, (const short*)gate_w, (const short*)up_w, (short*)h,
      sub_expert, sub_off, sub_cnt, sorted_tok, H, I);
}

extern "C" void launch_moe_down_grouped(const void* h, const void* down_w,
                                        float* out, const int* sub_expert,
                                        const int* sub_off, const int* sub_cnt,
                                        const int* sorted_tok,
                                        const float* sorted_wt, int S, int I,
                                        int H, hipStream_t stream) {
  size_t smem = (size_t)MG_TOK * I * sizeof(short);
  int gx = (H + MG_WAVES - 1) / MG_WAVES;
  if (gx > 64) gx = 64;
  moe_down_grouped_kernel<<<dim3(gx, S), dim3(MG_BLOCK), smem, stream>>>(
      (const short*)h, (const short*)down_w, out, sub_expert, sub_off,
      sub_cnt, sorted_tok, sorted_wt, I, H);
}

// ---------------------------------------------------------------------------
// MFMA bf16 grouped expert kernels (16-token sub-ranges).
//
// The scalar dot2 kernels above pay per-token VALU for every weight
// byte, which caps the profitable sub-range width at 4 tokens — expert
// weights get re-read ~1.5x at batch 64.  On MFMA the per-token cost is
// in the matrix unit, so 16-token sub-ranges stream each activated
// expert's weights ~once:
//   A = 16 weight rows x 32 k (bf16 16 B/lane loads — the row layout IS
//       the fragment layout), B = x^T from LDS, D[row, token].
// Same fragment scheme as the w4 MFMA kernel (w4a16.hip) minus dequant.
// ---------------------------------------------------------------------------

#define MF_WAVES 4
#define MF_BLOCK (MF_WAVES * WAVE)
#define MF_TOK 16
#define MF_CH 384
#define MF_LDS (MF_CH + 16)  // rows 16B-aligned, 8-bank shift per row
#define MF_NSL (MF_CH / 32)

typedef __bf16 mfbf16x8 __attribute__((ext_vector_type(8)));
typedef float mff32x4 __attribute__((ext_vector_type(4)));

// No LDS, no barriers: B-fragments are read straight from the 16 token
// rows in global (same strided 16-rows-x-16B pattern as the A loads —
// quarter-wave groups cover a full 64 B line per row).  x is tiny and
// L2-resident; staging it through LDS cost two barriers per chunk that
// serialized staging against compute.  Columns of dead tokens (t >=
// cnt) compute garbage that the epilogue simply never writes, so the
// token-row index is clamped instead of zero-padded.
__global__ __launch_bounds__(MF_BLOCK) void moe_gateup_mfma_kernel(
    const short* __restrict__ x,        // [N, H]
    const short* __restrict__ gate_w,   // [E, I, H]
    const short* __restrict__ up_w,     // [E, I, H]
    short* __restrict__ h,              // [P, I]
    const int* __restrict__ sub_expert, const int* __restrict__ sub_off,
    const int* __restrict__ sub_cnt, const int* __restrict__ sorted_tok,
    int H, int I) {
  const int s = blockIdx.y;
  const int e = sub_expert[s];
  const int p0 = sub_off[s];
  const int cnt = sub_cnt[s];
  if (cnt == 0) return;  // padded slot
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;

  const int row0 = (blockIdx.x * MF_WAVES + wid) * 16;
  if (row0 >= I) return;
  const long ebase = (long)e * I * H;
  const int wr = min(row0 + (lane & 15), I - 1);
  const short* grow = gate_w + ebase + (long)wr * H;
  const short* urow = up_w + ebase + (long)wr * H;
  const int tok_r = sorted_tok[p0 + min(lane & 15, cnt - 1)];
  const short* xrow = x + (long)tok_r * H;

  mff32x4 gacc = {0, 0, 0, 0}, uacc = {0, 0, 0, 0};
  // full MF_NSL-slice batches (all global loads issued before the MFMA
  // block — guide trap 4(b)), then an exact tail loop: without zeroed
  // LDS columns, clamped duplicate loads would double-count.
  const int nsl_total = H / 32;  // binding requires H % 32 == 0
  int sl = 0;
  for (; sl + MF_NSL <= nsl_total; sl += MF_NSL) {
    mfbf16x8 ga[MF_NSL], ua[MF_NSL], ba[MF_NSL];
#pragma unroll
    for (int i = 0; i < MF_NSL; ++i) {
      const int koff = (sl + i) * 32 + (lane >> 4) * 8;
      ga[i] = *reinterpret_cast<const mfbf16x8*>(grow + koff);
      ua[i] = *reinterpret_cast<const mfbf16x8*>(urow + koff);
      ba[i] = *reinterpret_cast<const mfbf16x8*>(xrow + koff);
    }
#pragma unroll
    for (int i = 0; i < MF_NSL; ++i) {
      gacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ga[i], ba[i], gacc, 0, 0, 0);
      uacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ua[i], ba[i], uacc, 0, 0, 0);
    }
  }
  for (; sl < nsl_total; ++sl) {  // < MF_NSL leftover slices, once
    const int koff = sl * 32 + (lane >> 4) * 8;
    mfbf16x8 ga = *reinterpret_cast<const mfbf16x8*>(grow + koff);
    mfbf16x8 ua = *reinterpret_cast<const mfbf16x8*>(urow + koff);
    mfbf16x8 ba = *reinterpret_cast<const mfbf16x8*>(xrow + koff);
    gacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ga, ba, gacc, 0, 0, 0);
    uacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ua, ba, uacc, 0, 0, 0);
  }

  const int tok = lane & 15;
  if (tok < cnt) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int o = row0 + (lane >> 4) * 4 + reg;
      if (o < I) {
        const float g = gacc[reg], u = uacc[reg];
        const float a = g / (1.0f + __expf(-g));  // silu
        h[(long)(p0 + tok) * I + o] = (short)__bfloat16_as_ushort(f2bf(a * u));
      }
    }
  }
}

__global__ __launch_bounds__(MF_BLOCK) void moe_down_mfma_kernel(
    const short* __restrict__ h,        // [P, I]
    const short* __restrict__ down_w,   // [E, H, I]
    float* __restrict__ out,            // [N, H] fp32 (pre-zeroed)
    const int* __restrict__ sub_expert, const int* __restrict__ sub_off,
    const int* __restrict__ sub_cnt, const int* __restrict__ sorted_tok,
    const float* __restrict__ sorted_wt, int I, int H) {
  const int s = blockIdx.y;
  const int e = sub_expert[s];
  const int p0 = sub_off[s];
  const int cnt = sub_cnt[s];
  if (cnt == 0) return;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;

  const int row0 = (blockIdx.x * MF_WAVES + wid) * 16;
  if (row0 >= H) return;
  const long ebase = (long)e * H * I;
  const int wr = min(row0 + (lane & 15), H - 1);
  const short* drow = down_w + ebase + (long)wr * I;
  // B rows = h pair rows (clamped for dead lanes; their D columns are
  // never written)
  const short* hrow = h + (long)(p0 + min(lane & 15, cnt - 1)) * I;

  mff32x4 acc = {0, 0, 0, 0};
  const int nsl_total = I / 32;  // binding requires I % 32 == 0
  int sl = 0;
  for (; sl + MF_NSL <= nsl_total; sl += MF_NSL) {
    mfbf16x8 da[MF_NSL], ba[MF_NSL];
#pragma unroll
    for (int i = 0; i < MF_NSL; ++i) {
      const int koff = (sl + i) * 32 + (lane >> 4) * 8;
      da[i] = *reinterpret_cast<const mfbf16x8*>(drow + koff);
      ba[i] = *reinterpret_cast<const mfbf16x8*>(hrow + koff);
    }
#pragma unroll
    for (int i = 0; i < MF_NSL; ++i)
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(da[i], ba[i], acc, 0, 0, 0);
  }
  for (; sl < nsl_total; ++sl) {
    const int koff = sl * 32 + (lane >> 4) * 8;
    mfbf16x8 da = *reinterpret_cast<const mfbf16x8*>(drow + koff);
    mfbf16x8 ba = *reinterpret_cast<const mfbf16x8*>(hrow + koff);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(da, ba, acc, 0, 0, 0);
  }

  const int tok = lane & 15;
  if (tok < cnt) {
    const float wt = sorted_wt[p0 + tok];
    const long trow = (long)sorted_tok[p0 + tok] * H;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int o = row0 + (lane >> 4) * 4 + reg;
      if (o < H) atomicAdd(out + trow + o, wt * acc[reg]);
    }
  }
}

extern "C" void launch_moe_gateup_mfma(const void* x, const void* gate_w,
                                       const void* up_w, void* h,
                                       const int* sub_expert,
                                       const int* sub_off, const int* sub_cnt,
                                       const int* sorted_tok, int S, int H,
                                       int I, hipStream_t stream) {
  const int gx = (I + MF_WAVES * 16 - 1) / (MF_WAVES * 16);
  moe_gateup_mfma_kernel<<<dim3(gx, S), dim3(MF_BLOCK), 0, stream>>>(
      (const short*)x, (const short*)gate_w, (const short*)up_w, (short*)h,
      sub_expert, sub_off, sub_cnt, sorted_tok, H, I);
}

extern "C" void launch_moe_down_mfma(const void* h, const void* down_w,
                                     float* out, const int* sub_expert,
                                     const int* sub_off, const int* sub_cnt,
                                     const int* sorted_tok,
                                     const float* sorted_wt, int S, int I,
                                     int H, hipStream_t stream) {
  const int gx = (H + MF_WAVES * 16 - 1) / (MF_WAVES * 16);
  moe_down_mfma_kernel<<<dim3(gx, S), dim3(MF_BLOCK), 0, stream>>>(
      (const short*)h, (const short*)down_w, out, sub_expert, sub_off,
      sub_cnt, sorted_tok, sorted_wt, I, H);
}

// ---------------------------------------------------------------------------
// Prefill padded-expert data movement.
//
// torch's advanced-indexing kernels for the [E, cap, H] scatter
// (`xp[dst] = x[tok]`) and the fp32 weighted `index_add_` reduce run
// ~5x off roofline at 16K-token prefill shapes; these two kernels do
// the same moves with vectorized row copies and a per-token K-way
// gather (fp32 accumulation, no atomics).
// ---------------------------------------------------------------------------

__global__ void moe_scatter_rows_kernel(const short* __restrict__ src,
                                        short* __restrict__ dst,
                                        const int* __restrict__ src_idx,
                                        const int* __restrict__ dst_idx,
                                        int H) {
  const int p = blockIdx.x;
  const short4v* s =
      reinterpret_cast<const short4v*>(src + (long)src_idx[p] * H);
  short4v* d = reinterpret_cast<short4v*>(dst + (long)dst_idx[p] * H);
  for (int i = threadIdx.x; i < H / 4; i += blockDim.x) d[i] = s[i];
}

// out[t] = sum_k wts[t,k] * d[pos[t,k]]  (bf16 rows, fp32 accumulate;
// short4 vector loads — the scalar form measured 2.3 TB/s on ~1 GB of
// traffic per prefill MoE layer)
__global__ void moe_gather_reduce_kernel(const short* __restrict__ d,
                                         const int* __restrict__ pos,    // [N,K]
                                         const float* __restrict__ wts,  // [N,K]
                                         short* __restrict__ out,        // [N,H]
                                         int K, int H) {
  const int t = blockIdx.x;
  const int* prow = pos + (long)t * K;
  const float* wrow = wts + (long)t * K;
  short* orow = out + (long)t * H;
  const int n4 = H / 4;
  for (int i = threadIdx.x; i < n4; i += blockDim.x) {
    float a0 = 0.0f, a1 = 0.0f, a2 = 0.0f, a3 = 0.0f;
    for (int k = 0; k < K; ++k) {
      const float w = wrow[k];
      short4v v = reinterpret_cast<const short4v*>(d + (long)prow[k] * H)[i];
      a0 += w * bfbits2f(v.x);
      a1 += w * bfbits2f(v.y);
      a2 += w * bfbits2f(v.z);
      a3 += w * bfbits2f(v.w);
    }
    short4v o;
    o.x = (short)__bfloat16_as_ushort(f2bf(a0));
    o.y = (short)__bfloat16_as_ushort(f2bf(a1));
    o.z = (short)__bfloat16_as_ushort(f2bf(a2));
    o.w = (short)__bfloat16_as_ushort(f2bf(a3));
    reinterpret_cast<short4v*>(orow)[i] = o;
  }
}

extern "C" void launch_moe_scatter_rows(const void* src, void* dst,
                                        const int* src_idx, const int* dst_idx,
                                        int P, int H, hipStream_t stream) {
  moe_scatter_rows_kernel<<<dim3((unsigned)P), dim3(256), 0, stream>>>(
      (const short*)src, (short*)dst, src_idx, dst_idx, H);
}

extern "C" void launch_moe_gather_reduce(const void* d, const int* pos,
                                         const float* wts, void* out, int N,
                                         int K, int H, hipStream_t stream) {
  moe_gather_reduce_kernel<<<dim3((unsigned)N), dim3(256), 0, stream>>>(
      (const short*)d, pos, wts, (short*)out, K, H);
}

// ---------------------------------------------------------------------------
// Fused MoE gating + expert sort + sub-range build (decode regime).
//
// Replaces ~20 small torch launches per MoE layer (softmax, topk,
// argsort/radix-sort, bincount, cumsums, scatter/cummax) with ONE
// kernel.  Greedy softmax top-k (DeepSeek-V2-Lite's topk_method);
// group-limited gating falls back to the torch path from Python.
//
// Single workgroup; one wave per token batch-slice; E <= 64 experts map
// one-per-lane.  N <= 64 tokens, top_k <= 8.
// ---------------------------------------------------------------------------

#define GK_MAXK 8
#define GK_MAXN 128  // max tokens (LDS arrays below)

// 1024 threads (16 waves): the top-k argmax is a serial chain of
// dependent ds_bpermute shuffles per token — more waves = fewer tokens
// serialized per wave.
__global__ __launch_bounds__(1024) void moe_gate_subranges_kernel(
    const short* __restrict__ logits,  // [N, E] bf16
    int* __restrict__ sorted_tok,      // [P]
    float* __restrict__ sorted_wt,     // [P]
    int* __restrict__ sub_expert,      // [s_upper]
    int* __restrict__ sub_off,         // [s_upper]
    int* __restrict__ sub_cnt,         // [s_upper]
    int N, int E, int K, int s_upper, int max_tok, float routed_scaling,
    int norm_topk) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;

  __shared__ int counts[64];        // per-expert pair count
  __shared__ int starts[64 + 1];    // exclusive prefix
  __shared__ int fill[64];          // scatter cursor per expert
  __shared__ int tok_e[GK_MAXN * GK_MAXK];    // chosen expert per (token, k)
  __shared__ float tok_w[GK_MAXN * GK_MAXK];  // chosen weight per (token, k)

  for (int i = threadIdx.x; i < 64; i += blockDim.x) {
    counts[i] = 0;
    fill[i] = 0;
  }
  __syncthreads();

  // ---- per-token softmax + greedy top-k (one wave per token) ----
  for (int t = wid; t < N; t += nw) {
    float sc = (lane < E) ? bfbits2f(logits[(long)t * E + lane]) : -1e30f;
    float mx = wave_max(sc);
    float p = (lane < E) ? __expf(sc - mx) : 0.0f;
    float denom = wave_sum(p);
    p /= denom;
    float psel = p;
    float wsum = 0.0f;
#pragma unroll
    for (int k = 0; k < GK_MAXK; ++k) {
      if (k >= K) break;
      // wave argmax over psel
      float best = psel;
      int bidx = lane;
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        float ov = __shfl_xor(best, off, WAVE);
        int oi = __shfl_xor(bidx, off, WAVE);
        if (ov > best || (ov == best && oi < bidx)) { best = ov; bidx = oi; }
      }
      if (lane == 0) {
        tok_e[t * GK_MAXK + k] = bidx;
        tok_w[t * GK_MAXK + k] = best;
        atomicAdd(&counts[bidx], 1);
      }
      wsum += best;
      if (lane == bidx) psel = -1.0f;
    }
    if (norm_topk && lane == 0) {
#pragma unroll
      for (int k = 0; k < GK_MAXK; ++k) {
        if (k >= K) break;
        tok_w[t * GK_MAXK + k] /= (wsum + 1e-20f);
      }
    }
  }
  __syncthreads();

  // ---- prefix sums (single wave) ----
  if (threadIdx.x == 0) {
    int acc = 0;
    for (int e = 0; e < E; ++e) {
      starts[e] = acc;
      acc += counts[e];
    }
    starts[E] = acc;
  }
  __syncthreads();

  // ---- scatter pairs into expert-sorted order ----
  for (int i = threadIdx.x; i < N * K; i += blockDim.x) {
    const int t = i / K, k = i % K;
    const int e = tok_e[t * GK_MAXK + k];
    const int pos = starts[e] + atomicAdd(&fill[e], 1);
    sorted_tok[pos] = t;
    sorted_wt[pos] = tok_w[t * GK_MAXK + k] * routed_scaling;
  }

  // ---- sub-range arrays ----
  // slot layout: expert e's ceil(counts[e]/max_tok) sub-ranges are
  // contiguous; one thread walks experts to assign slots (E <= 64, cheap)
  __shared__ int sub_base[64 + 1];
  if (threadIdx.x == 0) {
    int s = 0;
    for (int e = 0; e < E; ++e) {
      sub_base[e] = s;
      s += (counts[e] + max_tok - 1) / max_tok;
    }
    sub_base[E] = s;
  }
  __syncthreads();
  const int S = sub_base[E];
  for (int i = threadIdx.x; i < s_upper; i += blockDim.x) {
    if (i >= S) {
      sub_cnt[i] = 0;
      sub_expert[i] = 0;
      sub_off[i] = 0;
    }
  }
  for (int e = threadIdx.x; e < E; e += blockDim.x) {
    const int ns = (counts[e] + max_tok - 1) / max_tok;
    for (int j = 0; j < ns; ++j) {
      const int slot = sub_base[e] + j;
      sub_expert[slot] = e;
      sub_off[slot] = starts[e] + j * max_tok;
      sub_cnt[slot] = min(max_tok, counts[e] - j * max_tok);
    }
  }
}

extern "C" void launch_moe_gate_subranges(
    const void* logits, int* sorted_tok, float* sorted_wt, int* sub_expert,
    int* sub_off, int* sub_cnt, int N, int E, int K, int s_upper, int max_tok,
    float routed_scaling, int norm_topk, hipStream_t stream) {
  moe_gate_subranges_kernel<<<dim3(1), dim3(1024), 0, stream>>>(
      (const short*)logits, sorted_tok, sorted_wt, sub_expert, sub_off,
      sub_cnt, N, E, K, s_upper, max_tok, routed_scaling, norm_topk);
}

// ---------------------------------------------------------------------------
// MFMA w4 grouped-expert GEMM: the w4a16_mfma structure (w4a16.hip) with
// expert gather.  Sub-ranges carry up to 32 tokens (max_tok=32 gating);
// idle MFMA columns for small cnt cost nothing — the win is dequantizing
// each weight ONCE regardless of token count (the VALU-bound scalar
// kernel paid ~38 VALU per word PER TOKEN).
// ---------------------------------------------------------------------------

typedef __bf16 mw4bf16x8 __attribute__((ext_vector_type(8)));
typedef float mw4f32x4 __attribute__((ext_vector_type(4)));

#define MW_WAVES 4
#define MW_BLOCK (MW_WAVES * WAVE)
#define MW_MTOK 32
#define MW_NSL 8

// LDS-free (as the bf16 MFMA kernels above): the two B-fragment sets
// (tokens 0-15 / 16-31) are read straight from the L2-resident token
// rows — no staging barriers.  Dead-token columns (t >= cnt) read a
// clamped row and their outputs are never written.
template <int BITS>
__global__ __launch_bounds__(MW_BLOCK) void moe_w4_mfma_kernel(
    const short* __restrict__ x,          // [N, H] (or h [P, I])
    const unsigned int* __restrict__ wq,  // [E, O, H*BITS/32]
    const short* __restrict__ scales, const short* __restrict__ biases,
    short* __restrict__ y,                // [P, O] sorted pair order
    const int* __restrict__ sub_expert, const int* __restrict__ sub_off,
    const int* __restrict__ sub_cnt, const int* __restrict__ sorted_tok,
    int H, int O, int gs) {
  constexpr unsigned MASK = (1u << BITS) - 1u;
  const int s = blockIdx.y;
  const int e = sub_expert[s];
  const int p0 = sub_off[s];
  const int cnt = sub_cnt[s];
  if (cnt == 0) return;  // padded slot
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int words_per_row = H / (32 / BITS);

  const int row0 = (blockIdx.x * MW_WAVES + wid) * 16;
  if (row0 >= O) return;
  // two accumulators per output half, alternated by slice parity: PMC
  // showed 60% of wave cycles as ISSUE stalls (SQ_WAIT_INST_ANY) —
  // back-to-back MFMAs on one accumulator serialize on the dependent-
  // accumulator latency; parity-split chains double the dependency
  // distance.  Summed in the epilogue.
  mw4f32x4 acc0 = {0, 0, 0, 0}, acc1 = {0, 0, 0, 0};
  mw4f32x4 acc0b = {0, 0, 0, 0}, acc1b = {0, 0, 0, 0};
  const int wrow_r = min(row0 + (lane & 15), O - 1);
  const long ebase = (long)e * O;
  const unsigned int* wrow = wq + (ebase + wrow_r) * words_per_row;
  const short* srow = scales + (ebase + wrow_r) * (H / gs);
  const short* brow = biases + (ebase + wrow_r) * (H / gs);
  const short* xr0 = x + (long)sorted_tok[p0 + min(lane & 15, cnt - 1)] * H;
  const short* xr1 =
      x + (long)sorted_tok[p0 + min(16 + (lane & 15), cnt - 1)] * H;

  const int nsl_total = H / 32;  // binding requires H % 32 == 0
  int sl = 0;
  for (; sl + MW_NSL <= nsl_total; sl += MW_NSL) {
    unsigned int wbuf[MW_NSL * (BITS == 4 ? 1 : 2)];
    short sraw[MW_NSL], braw[MW_NSL];  // raw bf16: converting at
                                       // load-site forces a vmcnt wait
    mw4bf16x8 b0v[MW_NSL], b1v[MW_NSL];
#pragma unroll
    for (int i = 0; i < MW_NSL; ++i) {
      const int kk = (sl + i) * 32 + (lane >> 4) * 8;
      if (BITS == 4) {
        wbuf[i] = wrow[kk / 8];
      } else {
        wbuf[i * 2] = wrow[kk / 4];
        wbuf[i * 2 + 1] = wrow[kk / 4 + 1];
      }
      sraw[i] = srow[kk / gs];
      braw[i] = brow[kk / gs];
      b0v[i] = *reinterpret_cast<const mw4bf16x8*>(xr0 + kk);
      b1v[i] = *reinterpret_cast<const mw4bf16x8*>(xr1 + kk);
    }
    // dequant the WHOLE batch into registers first: a VALU-built A
    // fragment feeding its MFMA directly is a RAW issue-stall
    mw4bf16x8 afv[MW_NSL];
#pragma unroll
    for (int i = 0; i < MW_NSL; ++i) {
      const float sg = bfbits2f(sraw[i]);
      const float bg = bfbits2f(braw[i]);
      if (BITS == 4) {
        const unsigned int bits = wbuf[i];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          afv[i][j] = (__bf16)(sg * (float)((bits >> (4 * j)) & MASK) + bg);
      } else {
        const unsigned int b0 = wbuf[i * 2];
        const unsigned int b1 = wbuf[i * 2 + 1];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          afv[i][j] = (__bf16)(sg * (float)((b0 >> (8 * j)) & MASK) + bg);
          afv[i][4 + j] = (__bf16)(sg * (float)((b1 >> (8 * j)) & MASK) + bg);
        }
      }
    }
#pragma unroll
    for (int i = 0; i < MW_NSL; i += 2) {
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afv[i], b0v[i], acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afv[i], b1v[i], acc1, 0, 0, 0);
      acc0b = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afv[i + 1], b0v[i + 1],
                                                      acc0b, 0, 0, 0);
      acc1b = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afv[i + 1], b1v[i + 1],
                                                      acc1b, 0, 0, 0);
    }
  }
  for (; sl < nsl_total; ++sl) {  // < MW_NSL leftover slices, once
    const int kk = sl * 32 + (lane >> 4) * 8;
    const float sg = bfbits2f(srow[kk / gs]);
    const float bg = bfbits2f(brow[kk / gs]);
    mw4bf16x8 af;
    if (BITS == 4) {
      const unsigned int bits = wrow[kk / 8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        af[j] = (__bf16)(sg * (float)((bits >> (4 * j)) & MASK) + bg);
    } else {
      const unsigned int b0 = wrow[kk / 4];
      const unsigned int b1 = wrow[kk / 4 + 1];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        af[j] = (__bf16)(sg * (float)((b0 >> (8 * j)) & MASK) + bg);
        af[4 + j] = (__bf16)(sg * (float)((b1 >> (8 * j)) & MASK) + bg);
      }
    }
    mw4bf16x8 b0f = *reinterpret_cast<const mw4bf16x8*>(xr0 + kk);
    mw4bf16x8 b1f = *reinterpret_cast<const mw4bf16x8*>(xr1 + kk);
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b0f, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b1f, acc1, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    acc0[r] += acc0b[r];
    acc1[r] += acc1b[r];
  }

#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int o = row0 + (lane >> 4) * 4 + reg;
    if (o < O) {
      const int t0 = lane & 15;
      if (t0 < cnt)
        y[(long)(p0 + t0) * O + o] = (short)__bfloat16_as_ushort(f2bf(acc0[reg]));
      if (t0 + 16 < cnt)
        y[(long)(p0 + t0 + 16) * O + o] =
            (short)__bfloat16_as_ushort(f2bf(acc1[reg]));
    }
  }
}

extern "C" void launch_moe_w4_mfma(const void* x, const void* wq,
                                   const void* scales, const void* biases,
                                   void* y, const int* sub_expert,
                                   const int* sub_off, const int* sub_cnt,
                                   const int* sorted_tok, int S, int H, int O,
                                   int gs, int bits, hipStream_t stream) {
  const int gx = (O + MW_WAVES * 16 - 1) / (MW_WAVES * 16);
  if (bits == 4)
    moe_w4_mfma_kernel<4><<<dim3(gx, S), dim3(MW_BLOCK), 0, stream>>>(
        (const short*)x, (const unsigned int*)wq, (const short*)scales,
        (const short*)biases, (short*)y, sub_expert, sub_off, sub_cnt,
        sorted_tok, H, O, gs);
  else
    moe_w4_mfma_kernel<8><<<dim3(gx, S), dim3(MW_BLOCK), 0, stream>>>(
        (const short*)x, (const unsigned int*)wq, (const short*)scales,
        (const short*)biases, (short*)y, sub_expert, sub_off, sub_cnt,
        sorted_tok, H, O, gs);
}
