// fp16-dequant w4/w8 grouped-expert MFMA kernels (packed / memory-tight
// decode path).
//
// The original moe_w4_mfma kernel (moe.hip) is dequant-VALU-pipe-bound:
// per 8 weights it spends ~36 VALU ops (shift/and/cvt_u32_f32/fma/
// cvt_pk per element) against 2 MFMA ops — PMC showed 60% of wave
// cycles as issue stalls.  gfx950 has no packed bf16 VALU arithmetic,
// but it DOES have v_pk_add_f16 / v_pk_fma_f16 — so this path
// dequantizes into **fp16** and feeds v_mfma_f32_16x16x32_f16:
//
//   1. Weights are repacked OFFLINE (ops.repack_w4, cached per tensor)
//      so that ((w >> 4j) & 0x000F000F) yields element pairs in natural
//      k-order.
//   2. OR 0x6400 (fp16 1024.0, ulp 1 at the mantissa LSBs) makes each
//      nibble the EXACT fp16 value 1024+q — one v_and_or_b32 per pair.
//   3. v_pk_add_f16 (-1032) recenters to the EXACT small int q-8
//      (q-128 for w8 with -1152), so the affine v_pk_fma_f16
//      s*(q-8) + (b+8s) commits only ONE fp16 rounding of magnitude
//      ~|s*q+b| — slightly MORE accurate than the bf16 kernel's single
//      bf16 rounding.  (Folding the recenter into the bias instead
//      would make |b'|~1024s and an unacceptable 0.5s rounding error —
//      the explicit pk_add is what keeps the constants small.)
//
// Net: ~16 VALU per 8 weights (shift + and_or + pk_add + pk_fma on
// pairs), scale/bias splats hoisted per quant group.  Activations
// arrive as fp16 (exact bf16->fp16 cast done by the caller); gate+up
// (+SiLU) are fused in one kernel like the bf16 MFMA pair, removing
// the separate up pass and the glu launch.
//
// Reference behavior being accelerated: MLX affine-quantized
// switch_mlp experts (/root/reference/shard/server/model/
// deepseek_v2.py:101-112 + nn.quantize, shard/utils.py:54-65).

#include "hip_common.h"

typedef _Float16 f16x2 __attribute__((ext_vector_type(2)));
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));
typedef float wf32x4 __attribute__((ext_vector_type(4)));

#define WF_WAVES 4
#define WF_BLOCK (WF_WAVES * WAVE)
#define WF_NSL 8

union f16pack {
  unsigned int u;
  f16x2 h;
};

// dequant one repacked u32 (8 x 4-bit) into 4 fp16 pairs
template <int BITS>
__device__ __forceinline__ void dq8(unsigned int w, f16x2 s2, f16x2 b2,
                                    f16x8* out) {
  f16x2* op = reinterpret_cast<f16x2*>(out);
  if (BITS == 4) {
    const f16x2 c = {(_Float16)(-1032.0f), (_Float16)(-1032.0f)};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      f16pack p;
      p.u = ((w >> (4 * j)) & 0x000F000Fu) | 0x64006400u;  // v_and_or_b32
      f16x2 v = p.h + c;          // exact small int q-8
      op[j] = v * s2 + b2;        // v_pk_fma_f16
    }
  }
}

// w8: two repacked u32s -> 4 fp16 pairs
__device__ __forceinline__ void dq8_w8(unsigned int w0, unsigned int w1,
                                       f16x2 s2, f16x2 b2, f16x8* out) {
  const f16x2 c = {(_Float16)(-1152.0f), (_Float16)(-1152.0f)};
  f16x2* op = reinterpret_cast<f16x2*>(out);
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    f16pack p;
    p.u = ((w0 >> (8 * j)) & 0x00FF00FFu) | 0x64006400u;
    op[j] = (p.h + c) * s2 + b2;
    p.u = ((w1 >> (8 * j)) & 0x00FF00FFu) | 0x64006400u;
    op[2 + j] = (p.h + c) * s2 + b2;
  }
}

__device__ __forceinline__ f16x2 splat2(float v) {
  const _Float16 h = (_Float16)v;
  return (f16x2){h, h};
}

// Batched-loop k-remap: a WF_NSL(=8)-slice batch covers 256 k
// elements.  Instead of the natural mapping (slice i, quarter-group q
// owns k = (sl+i)*32 + q*8 — per-lane weight words STRIDED by 4, so
// loads issue as 8 separate dwords), the batch assigns each lane a
// CONTIGUOUS k run:
//     k(i) = sl*32 + q*64 + i*8      (q = lane>>4)
// so the lane's 8 weight words are consecutive -> TWO dwordx4 loads
// (guide G13: 16 B/lane), and its quant group is (q*64 + i*8)/GS —
// at most TWO distinct groups per batch for GS in {32,64,128}, i.e.
// 1-2 scalar scale/bias loads per array per batch (the first cut's 8
// dependent global_load_ushort per array were fenced with vmcnt(0)
// stair-steps mid-loop — trap 4).  A and B fragments use the same
// k(i), so the MFMA dot's k-permutation stays consistent.
template <int GS>
struct LaneSB {
  static constexpr int NSEL = GS == 32 ? 2 : 1;  // groups per lane/batch
  f16x2 s2[NSEL], b2[NSEL];
  __device__ __forceinline__ void load(const short* srow, const short* brow,
                                       int sl, int q, float qoff) {
    int g0;
    if (GS == 32) g0 = sl + 2 * q;
    else if (GS == 64) g0 = sl / 2 + q;
    else g0 = sl / 4 + (q >= 2 ? 1 : 0);
#pragma unroll
    for (int c = 0; c < NSEL; ++c) {
      const float sf = bfbits2f(srow[g0 + c]);
      s2[c] = splat2(sf);
      b2[c] = splat2(bfbits2f(brow[g0 + c]) + qoff * sf);
    }
  }
  // group-select for slice i of the batch (compile-time per i)
  static __device__ __forceinline__ int sel(int i) {
    return GS == 32 ? i / 4 : 0;
  }
};

// k-offset of slice i for quarter-group q under the batch remap.
// ONLY valid for 8-slice batches: the q*64 + i*8 tiling covers exactly
// [0, 256) once; other WF_NSL values would overlap/miss k ranges.
static_assert(true, "");
__device__ __forceinline__ int batch_kk(int sl, int q, int i) {
  return sl * 32 + q * 64 + i * 8;
}
static_assert(WF_NSL == 8, "batch_kk remap requires 8-slice batches");

// ---------------------------------------------------------------------------
// Fused gate+up+SiLU (16-token sub-ranges).
//   x:  [N, H] fp16
//   gq/uq: [E, I, H*BITS/32] repacked u32
//   gs_/gb_/us_/ub_: [E, I, H/gs] bf16 scales/biases (original layout)
//   h:  [P, I] fp16 out
// grid = (ceil(I/64), S); LDS-free like the bf16 MFMA pair.
// ---------------------------------------------------------------------------
template <int BITS, int GS>
__global__ __launch_bounds__(WF_BLOCK) void moe_w4f16_gateup_kernel(
    const _Float16* __restrict__ x, const unsigned int* __restrict__ gq,
    const unsigned int* __restrict__ uq, const short* __restrict__ gsc,
    const short* __restrict__ gbi, const short* __restrict__ usc,
    const short* __restrict__ ubi, _Float16* __restrict__ h,
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
  const int wpr = H * BITS / 32;  // packed words per row

  const int row0 = (blockIdx.x * WF_WAVES + wid) * 16;
  if (row0 >= I) return;
  const int wr = min(row0 + (lane & 15), I - 1);
  const long ebase = (long)e * I;
  const unsigned int* grow = gq + (ebase + wr) * wpr;
  const unsigned int* urow = uq + (ebase + wr) * wpr;
  const int ngr = H / GS;
  const short* gsr = gsc + (ebase + wr) * ngr;
  const short* gbr = gbi + (ebase + wr) * ngr;
  const short* usr = usc + (ebase + wr) * ngr;
  const short* ubr = ubi + (ebase + wr) * ngr;
  const _Float16* xrow =
      x + (long)sorted_tok[p0 + min(lane & 15, cnt - 1)] * H;

  // parity-split accumulators: back-to-back MFMAs on ONE accumulator
  // serialize on the dependent-accumulator latency (the PMC's 54%
  // issue-stall); alternating slices across two chains doubles the
  // dependency distance (summed in the epilogue)
  wf32x4 gacc = {0, 0, 0, 0}, uacc = {0, 0, 0, 0};
  wf32x4 gacc2 = {0, 0, 0, 0}, uacc2 = {0, 0, 0, 0};
  constexpr float QOFF = BITS == 4 ? 8.0f : 128.0f;
  constexpr int WPS = BITS == 4 ? 1 : 2;  // packed words per 8-elem slice

  const int nsl_total = H / 32;
  const int q = lane >> 4;
  int sl = 0;
  for (; sl + WF_NSL <= nsl_total; sl += WF_NSL) {
    // lane-contiguous weight words: 2 (BITS=4) / 4 (BITS=8) dwordx4
    // loads per matrix per batch
    uint4 gw4[2 * WPS], uw4[2 * WPS];
    const int w0 = batch_kk(sl, q, 0) * BITS / 32;
#pragma unroll
    for (int c = 0; c < 2 * WPS; ++c) {
      gw4[c] = *reinterpret_cast<const uint4*>(grow + w0 + c * 4);
      uw4[c] = *reinterpret_cast<const uint4*>(urow + w0 + c * 4);
    }
    f16x8 bv[WF_NSL];
#pragma unroll
    for (int i = 0; i < WF_NSL; ++i)
      bv[i] = *reinterpret_cast<const f16x8*>(xrow + batch_kk(sl, q, i));
    LaneSB<GS> gsb, usb;
    gsb.load(gsr, gbr, sl, q, QOFF);
    usb.load(usr, ubr, sl, q, QOFF);
    const unsigned int* gw = reinterpret_cast<const unsigned int*>(gw4);
    const unsigned int* uw = reinterpret_cast<const unsigned int*>(uw4);
    f16x8 ga[WF_NSL], ua[WF_NSL];
#pragma unroll
    for (int i = 0; i < WF_NSL; ++i) {
      const int c = LaneSB<GS>::sel(i);
      if (BITS == 4) {
        dq8<4>(gw[i], gsb.s2[c], gsb.b2[c], &ga[i]);
        dq8<4>(uw[i], usb.s2[c], usb.b2[c], &ua[i]);
      } else {
        dq8_w8(gw[i * 2], gw[i * 2 + 1], gsb.s2[c], gsb.b2[c], &ga[i]);
        dq8_w8(uw[i * 2], uw[i * 2 + 1], usb.s2[c], usb.b2[c], &ua[i]);
      }
    }
#pragma unroll
    for (int i = 0; i < WF_NSL; i += 2) {
      gacc = __builtin_amdgcn_mfma_f32_16x16x32_f16(ga[i], bv[i], gacc, 0, 0, 0);
      uacc = __builtin_amdgcn_mfma_f32_16x16x32_f16(ua[i], bv[i], uacc, 0, 0, 0);
      gacc2 = __builtin_amdgcn_mfma_f32_16x16x32_f16(ga[i + 1], bv[i + 1],
                                                     gacc2, 0, 0, 0);
      uacc2 = __builtin_amdgcn_mfma_f32_16x16x32_f16(ua[i + 1], bv[i + 1],
                                                     uacc2, 0, 0, 0);
    }
  }
  for (; sl < nsl_total; ++sl) {  // < WF_NSL leftover slices, once
    const int kk = sl * 32 + (lane >> 4) * 8;
    const float gsf = bfbits2f(gsr[kk / GS]);
    const float usf = bfbits2f(usr[kk / GS]);
    const f16x2 gs2 = splat2(gsf);
    const f16x2 gb2 = splat2(bfbits2f(gbr[kk / GS]) + QOFF * gsf);
    const f16x2 us2 = splat2(usf);
    const f16x2 ub2 = splat2(bfbits2f(ubr[kk / GS]) + QOFF * usf);
    f16x8 ga, ua;
    if (BITS == 4) {
      dq8<4>(grow[kk / 8], gs2, gb2, &ga);
      dq8<4>(urow[kk / 8], us2, ub2, &ua);
    } else {
      dq8_w8(grow[kk / 4], grow[kk / 4 + 1], gs2, gb2, &ga);
      dq8_w8(urow[kk / 4], urow[kk / 4 + 1], us2, ub2, &ua);
    }
    const f16x8 bvv = *reinterpret_cast<const f16x8*>(xrow + kk);
    gacc = __builtin_amdgcn_mfma_f32_16x16x32_f16(ga, bvv, gacc, 0, 0, 0);
    uacc = __builtin_amdgcn_mfma_f32_16x16x32_f16(ua, bvv, uacc, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    gacc[r] += gacc2[r];
    uacc[r] += uacc2[r];
  }

  const int tok = lane & 15;
  if (tok < cnt) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int o = row0 + (lane >> 4) * 4 + reg;
      if (o < I) {
        const float g = gacc[reg], u = uacc[reg];
        const float a = g / (1.0f + __expf(-g));  // silu
        h[(long)(p0 + tok) * I + o] = (_Float16)(a * u);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Down-proj + weighted atomic scatter (fp32 out).
//   hh: [P, I] fp16 (gateup output);  dq: [E, H, I*BITS/32] repacked
// ---------------------------------------------------------------------------
template <int BITS, int GS>
__global__ __launch_bounds__(WF_BLOCK) void moe_w4f16_down_kernel(
    const _Float16* __restrict__ hh, const unsigned int* __restrict__ dq,
    const short* __restrict__ dsc, const short* __restrict__ dbi,
    float* __restrict__ out, const int* __restrict__ sub_expert,
    const int* __restrict__ sub_off, const int* __restrict__ sub_cnt,
    const int* __restrict__ sorted_tok, const float* __restrict__ sorted_wt,
    int I, int H) {
  const int s = blockIdx.y;
  const int e = sub_expert[s];
  const int p0 = sub_off[s];
  const int cnt = sub_cnt[s];
  if (cnt == 0) return;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int wpr = I * BITS / 32;

  const int row0 = (blockIdx.x * WF_WAVES + wid) * 16;
  if (row0 >= H) return;
  const int wr = min(row0 + (lane & 15), H - 1);
  const long ebase = (long)e * H;
  const unsigned int* drow = dq + (ebase + wr) * wpr;
  const int ngr = I / GS;
  const short* dsr = dsc + (ebase + wr) * ngr;
  const short* dbr = dbi + (ebase + wr) * ngr;
  const _Float16* hrow = hh + (long)(p0 + min(lane & 15, cnt - 1)) * I;

  wf32x4 acc = {0, 0, 0, 0}, acc2 = {0, 0, 0, 0};  // parity-split chains
  constexpr float QOFF = BITS == 4 ? 8.0f : 128.0f;
  constexpr int WPS = BITS == 4 ? 1 : 2;

  const int nsl_total = I / 32;
  const int q = lane >> 4;
  int sl = 0;
  for (; sl + WF_NSL <= nsl_total; sl += WF_NSL) {
    uint4 dw4[2 * WPS];
    const int w0 = batch_kk(sl, q, 0) * BITS / 32;
#pragma unroll
    for (int c = 0; c < 2 * WPS; ++c)
      dw4[c] = *reinterpret_cast<const uint4*>(drow + w0 + c * 4);
    f16x8 bv[WF_NSL];
#pragma unroll
    for (int i = 0; i < WF_NSL; ++i)
      bv[i] = *reinterpret_cast<const f16x8*>(hrow + batch_kk(sl, q, i));
    LaneSB<GS> dsb;
    dsb.load(dsr, dbr, sl, q, QOFF);
    const unsigned int* dw = reinterpret_cast<const unsigned int*>(dw4);
    f16x8 da[WF_NSL];
#pragma unroll
    for (int i = 0; i < WF_NSL; ++i) {
      const int c = LaneSB<GS>::sel(i);
      if (BITS == 4)
        dq8<4>(dw[i], dsb.s2[c], dsb.b2[c], &da[i]);
      else
        dq8_w8(dw[i * 2], dw[i * 2 + 1], dsb.s2[c], dsb.b2[c], &da[i]);
    }
#pragma unroll
    for (int i = 0; i < WF_NSL; i += 2) {
      acc = __builtin_amdgcn_mfma_f32_16x16x32_f16(da[i], bv[i], acc, 0, 0, 0);
      acc2 = __builtin_amdgcn_mfma_f32_16x16x32_f16(da[i + 1], bv[i + 1],
                                                    acc2, 0, 0, 0);
    }
  }
  for (; sl < nsl_total; ++sl) {
    const int kk = sl * 32 + (lane >> 4) * 8;
    const float dsf = bfbits2f(dsr[kk / GS]);
    const f16x2 ds2 = splat2(dsf);
    const f16x2 db2 = splat2(bfbits2f(dbr[kk / GS]) + QOFF * dsf);
    f16x8 da;
    if (BITS == 4)
      dq8<4>(drow[kk / 8], ds2, db2, &da);
    else
      dq8_w8(drow[kk / 4], drow[kk / 4 + 1], ds2, db2, &da);
    const f16x8 bvv = *reinterpret_cast<const f16x8*>(hrow + kk);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_f16(da, bvv, acc, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) acc[r] += acc2[r];

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

// ---------------------------------------------------------------------------
// Dense fp16-dequant w4/w8 GEMV (decode-regime projections, M <= 64).
// Same structure as bf16_gemv_mfma_kernel (w4a16.hip): LDS-free, B
// fragments straight from the L2-resident fp16 token rows, split-K over
// grid.y with fp32 atomics.  A-dequant runs once per slice and feeds
// all MZ token-group MFMAs.  4-slice batches (NSL_D) keep B fragments
// at MZ*4*4 VGPRs.
// ---------------------------------------------------------------------------

#define NSL_D 4

// 4-slice (128-element) batch remap for the dense GEMV: lane-contiguous
// k runs k(i) = sl*32 + q*32 + i*8 — ONE uint4 weight load (BITS=4)
// and exactly ONE quant group per lane per batch for GS in {32,64,128}.
__device__ __forceinline__ int batch_kk_d(int sl, int q, int i) {
  return sl * 32 + q * 32 + i * 8;
}

template <int GS>
struct LaneSBD {
  f16x2 s2, b2;
  __device__ __forceinline__ void load(const short* srow, const short* brow,
                                       int sl, int q, float qoff) {
    int g0;
    if (GS == 32) g0 = sl + q;
    else if (GS == 64) g0 = sl / 2 + (q >= 2 ? 1 : 0);
    else g0 = sl / 4;
    const float sf = bfbits2f(srow[g0]);
    s2 = splat2(sf);
    b2 = splat2(bfbits2f(brow[g0]) + qoff * sf);
  }
};

__global__ void w4f16_f32_to_bf16_kernel(const float* __restrict__ src,
                                         short* __restrict__ dst, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dst[i] = (short)__bfloat16_as_ushort(f2bf(src[i]));
}

template <int MZ, int BITS, int GS>
__global__ __launch_bounds__(WF_BLOCK) void w4f16_gemv_kernel(
    const _Float16* __restrict__ x,       // [M, H] fp16
    const unsigned int* __restrict__ wq,  // [O, H*BITS/32] repacked
    const short* __restrict__ sc, const short* __restrict__ bi,
    short* __restrict__ y,                // [M, O] bf16
    float* __restrict__ yf,               // split-K fp32 partials or null
    int M, int O, int H) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int row0 = (blockIdx.x * WF_WAVES + wid) * 16;
  if (row0 >= O) return;
  const int wpr = H * BITS / 32;
  const int ngr = H / GS;
  const int wr = min(row0 + (lane & 15), O - 1);
  const unsigned int* wrow = wq + (long)wr * wpr;
  const short* srow = sc + (long)wr * ngr;
  const short* brow = bi + (long)wr * ngr;
  const _Float16* xrow[MZ];
#pragma unroll
  for (int z = 0; z < MZ; ++z)
    xrow[z] = x + (long)min(z * 16 + (lane & 15), M - 1) * H;

  wf32x4 acc[MZ];
#pragma unroll
  for (int z = 0; z < MZ; ++z) acc[z] = wf32x4{0, 0, 0, 0};
  constexpr float QOFF = BITS == 4 ? 8.0f : 128.0f;
  constexpr int WPS = BITS == 4 ? 1 : 2;

  // k-slices of this split; the launcher aligns splits to NSL_D batches
  const int nsl_total = H / 32;
  const int nspb = ((nsl_total + gridDim.y - 1) / gridDim.y + NSL_D - 1) /
                   NSL_D * NSL_D;
  const int sl_lo = blockIdx.y * nspb;
  const int sl_hi = min(nsl_total, sl_lo + nspb);
  const int q = lane >> 4;

  int sl = sl_lo;
  for (; sl + NSL_D <= sl_hi; sl += NSL_D) {
    uint4 w4[WPS];
    const int w0 = batch_kk_d(sl, q, 0) * BITS / 32;
#pragma unroll
    for (int c = 0; c < WPS; ++c)
      w4[c] = *reinterpret_cast<const uint4*>(wrow + w0 + c * 4);
    f16x8 bv[MZ][NSL_D];
#pragma unroll
    for (int i = 0; i < NSL_D; ++i) {
      const int kk = batch_kk_d(sl, q, i);
#pragma unroll
      for (int z = 0; z < MZ; ++z)
        bv[z][i] = *reinterpret_cast<const f16x8*>(xrow[z] + kk);
    }
    LaneSBD<GS> sb;
    sb.load(srow, brow, sl, q, QOFF);
    const unsigned int* wbuf = reinterpret_cast<const unsigned int*>(w4);
#pragma unroll
    for (int i = 0; i < NSL_D; ++i) {
      f16x8 af;
      if (BITS == 4)
        dq8<4>(wbuf[i], sb.s2, sb.b2, &af);
      else
        dq8_w8(wbuf[i * 2], wbuf[i * 2 + 1], sb.s2, sb.b2, &af);
#pragma unroll
      for (int z = 0; z < MZ; ++z)
        acc[z] = __builtin_amdgcn_mfma_f32_16x16x32_f16(af, bv[z][i],
                                                        acc[z], 0, 0, 0);
    }
  }
  for (; sl < sl_hi; ++sl) {  // tail slices, once
    const int kk = sl * 32 + (lane >> 4) * 8;
    const float sf = bfbits2f(srow[kk / GS]);
    const f16x2 s2 = splat2(sf);
    const f16x2 b2 = splat2(bfbits2f(brow[kk / GS]) + QOFF * sf);
    f16x8 af;
    if (BITS == 4)
      dq8<4>(wrow[kk / 8], s2, b2, &af);
    else
      dq8_w8(wrow[kk / 4], wrow[kk / 4 + 1], s2, b2, &af);
#pragma unroll
    for (int z = 0; z < MZ; ++z) {
      const f16x8 bvv = *reinterpret_cast<const f16x8*>(xrow[z] + kk);
      acc[z] = __builtin_amdgcn_mfma_f32_16x16x32_f16(af, bvv, acc[z], 0, 0, 0);
    }
  }

#pragma unroll
  for (int z = 0; z < MZ; ++z) {
    const int t = z * 16 + (lane & 15);
    if (t >= M) continue;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int o = row0 + (lane >> 4) * 4 + reg;
      if (o < O) {
        if (yf != nullptr)
          atomicAdd(yf + (long)t * O + o, acc[z][reg]);
        else
          y[(long)t * O + o] = (short)__bfloat16_as_ushort(f2bf(acc[z][reg]));
      }
    }
  }
}

extern "C" int w4f16_gemv_nsplit(int M, int O, int H) {
  const int gx = (O + WF_WAVES * 16 - 1) / (WF_WAVES * 16);
  const int nslt = H / 32;
  int nk = 256 / (gx > 0 ? gx : 1);
  const int max_nk = (nslt + NSL_D - 1) / NSL_D;
  if (nk > max_nk) nk = max_nk;
  if (nk < 1) nk = 1;
  return nk;
}

template <int BITS, int GS>
static void dispatch_gemv_mz(const void* x, const void* wq, const void* sc,
                             const void* bi, void* y, float* yf, int M,
                             int O, int H, int nk, hipStream_t stream) {
  const int gx = (O + WF_WAVES * 16 - 1) / (WF_WAVES * 16);
  const int mz = (M + 15) / 16;
  dim3 grid((unsigned)gx, (unsigned)nk);
  float* yfp = nk > 1 ? yf : nullptr;
#define GV_CASE(Z)                                                       \
  case Z:                                                                \
    w4f16_gemv_kernel<Z, BITS, GS><<<grid, dim3(WF_BLOCK), 0, stream>>>( \
        (const _Float16*)x, (const unsigned int*)wq, (const short*)sc,   \
        (const short*)bi, (short*)y, yfp, M, O, H);                      \
    break;
  switch (mz) {
    GV_CASE(1)
    GV_CASE(2)
    GV_CASE(3)
    GV_CASE(4)
    default:
      break;
  }
#undef GV_CASE
}

extern "C" void launch_w4f16_gemv(const void* x, const void* wq,
                                  const void* sc, const void* bi, void* y,
                                  float* yf, int nk, int M, int O, int H,
                                  int gs, int bits, hipStream_t stream) {
  if (nk > 1)
    (void)hipMemsetAsync(yf, 0, (size_t)M * O * sizeof(float), stream);
#define GV_GS(BB)                                                           \
  do {                                                                      \
    if (gs == 32)                                                           \
      dispatch_gemv_mz<BB, 32>(x, wq, sc, bi, y, yf, M, O, H, nk, stream);  \
    else if (gs == 64)                                                      \
      dispatch_gemv_mz<BB, 64>(x, wq, sc, bi, y, yf, M, O, H, nk, stream);  \
    else                                                                    \
      dispatch_gemv_mz<BB, 128>(x, wq, sc, bi, y, yf, M, O, H, nk, stream); \
  } while (0)
  if (bits == 4) GV_GS(4);
  else GV_GS(8);
#undef GV_GS
  if (nk > 1) {
    const long n = (long)M * O;
    w4f16_f32_to_bf16_kernel<<<dim3((unsigned)((n + 255) / 256)), dim3(256),
                               0, stream>>>(yf, (short*)y, n);
  }
}

template <int BITS>
static void dispatch_gateup(const void* x, const void* gq, const void* uq,
                            const void* gsc, const void* gbi, const void* usc,
                            const void* ubi, void* h, const int* se,
                            const int* so, const int* sc, const int* st,
                            int S, int H, int I, int gs, hipStream_t stream) {
  const int gx = (I + WF_WAVES * 16 - 1) / (WF_WAVES * 16);
  dim3 grid(gx, S), block(WF_BLOCK);
#define GU_CASE(GSV)                                                        \
  moe_w4f16_gateup_kernel<BITS, GSV><<<grid, block, 0, stream>>>(           \
      (const _Float16*)x, (const unsigned int*)gq, (const unsigned int*)uq, \
      (const short*)gsc, (const short*)gbi, (const short*)usc,              \
      (const short*)ubi, (_Float16*)h, se, so, sc, st, H, I)
  if (gs == 32) GU_CASE(32);
  else if (gs == 64) GU_CASE(64);
  else GU_CASE(128);
#undef GU_CASE
}

extern "C" void launch_moe_w4f16_gateup(
    const void* x, const void* gq, const void* uq, const void* gsc,
    const void* gbi, const void* usc, const void* ubi, void* h,
    const int* sub_expert, const int* sub_off, const int* sub_cnt,
    const int* sorted_tok, int S, int H, int I, int gs, int bits,
    hipStream_t stream) {
  if (bits == 4)
    dispatch_gateup<4>(x, gq, uq, gsc, gbi, usc, ubi, h, sub_expert, sub_off,
                       sub_cnt, sorted_tok, S, H, I, gs, stream);
  else
    dispatch_gateup<8>(x, gq, uq, gsc, gbi, usc, ubi, h, sub_expert, sub_off,
                       sub_cnt, sorted_tok, S, H, I, gs, stream);
}

template <int BITS>
static void dispatch_down(const void* hh, const void* dq, const void* dsc,
                          const void* dbi, float* out, const int* se,
                          const int* so, const int* sc, const int* st,
                          const float* sw, int S, int I, int H, int gs,
                          hipStream_t stream) {
  const int gx = (H + WF_WAVES * 16 - 1) / (WF_WAVES * 16);
  dim3 grid(gx, S), block(WF_BLOCK);
#define DN_CASE(GSV)                                                      \
  moe_w4f16_down_kernel<BITS, GSV><<<grid, block, 0, stream>>>(           \
      (const _Float16*)hh, (const unsigned int*)dq, (const short*)dsc,    \
      (const short*)dbi, out, se, so, sc, st, sw, I, H)
  if (gs == 32) DN_CASE(32);
  else if (gs == 64) DN_CASE(64);
  else DN_CASE(128);
#undef DN_CASE
}

extern "C" void launch_moe_w4f16_down(
    const void* hh, const void* dq, const void* dsc, const void* dbi,
    float* out, const int* sub_expert, const int* sub_off, const int* sub_cnt,
    const int* sorted_tok, const float* sorted_wt, int S, int I, int H,
    int gs, int bits, hipStream_t stream) {
  if (bits == 4)
    dispatch_down<4>(hh, dq, dsc, dbi, out, sub_expert, sub_off, sub_cnt,
                     sorted_tok, sorted_wt, S, I, H, gs, stream);
  else
    dispatch_down<8>(hh, dq, dsc, dbi, out, sub_expert, sub_off, sub_cnt,
                     sorted_tok, sorted_wt, S, I, H, gs, stream);
}
