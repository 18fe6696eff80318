// Flash-decode attention (T=1) over the KV cache — the decode-latency
// hot op (SURVEY.md §7 hard-part #2).
//
// Shapes: q [B, Hq, 1, Dk], kcache [B, Hkv, Scap, Dk], vcache
// [B, Hkv, Scap, Dv]; S = valid length (host int or device pos+1 for
// hipGraph capture).  GQA: one block serves one (batch, kv-head) pair
// and computes all G = Hq/Hkv query heads at once, so each K/V byte is
// read once regardless of the GQA ratio.  Supports MLA shapes (Dk=192,
// Dv=128), gemma2 softcap + sliding window.  fp32 accumulation, online
// softmax over key tiles.
//
// Split-S: at B*Hkv blocks the chip is parallelism-starved (512 blocks
// on 256 CUs), so the key range is split over gridDim.y slices, each
// writing unnormalized partials (m, l, O·l) that a tiny combine kernel
// reduces — flash-decode split-K.  Slice ranges are fractions of the
// CAPACITY so they are static under graph capture; slices past S
// produce (m=-inf, l=0) partials the combiner ignores.
//
// G is a template parameter so per-head register arrays stay statically
// indexed (guide §5.4 rule 20).

#include "hip_common.h"

#define AD_BLOCK 256

// DVT = v-columns per thread (ceil(Dv / AD_BLOCK)): 1 for Dv <= 256,
// 2 for the absorbed-MLA Dv = 512.  vstride is the per-key element
// stride of V rows — for absorbed MLA, V is the first kv_lora_rank
// columns of the K rows themselves (vstride = Dk, vcache = kcache).
template <int G, int DVT>
__global__ __launch_bounds__(AD_BLOCK) void attn_decode_kernel(
    const short* __restrict__ q,      // [B, Hq, Dk]
    const short* __restrict__ kcache, // [B, Hkv, Scap, Dk]
    const short* __restrict__ vcache, // [B, Hkv, Scap, *] (vstride elems/key)
    short* __restrict__ out,          // [B, Hq, Dv]      (nsplit == 1)
    float* __restrict__ part_o,       // [B, Hkv, G, NS, Dv]  (nsplit > 1)
    float* __restrict__ part_ml,      // [B, Hkv, G, NS, 2]
    const int* __restrict__ s_ptr,    // device position (S = *s_ptr + 1), or null
    int B, int Hq, int Hkv, int S, long Scap, int Dk, int Dv, long vstride,
    float scale, float softcap, int window) {
  if (s_ptr) S = *s_ptr + 1;  // hipGraph-captured decode: length lives on device
  const int b = blockIdx.x / Hkv;
  const int hk = blockIdx.x % Hkv;
  const int split = blockIdx.y;
  const int nsplit = gridDim.y;
  // Head-group split (gridDim.z): big-G shapes (absorbed MLA: 16 heads
  // over one KV "head") run as NHG blocks of G heads each — smaller
  // per-block register state → real occupancy; the re-read K tile
  // stays L2-resident.  head id = hk*Gtot + hg*G + g.
  const int hg = blockIdx.z;
  const int Gtot = (Hq / Hkv);
  const int h0 = hk * Gtot + hg * G;
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  short* q_lds = reinterpret_cast<short*>(smem_raw);           // [G][Dk] bf16
  float* p_lds = reinterpret_cast<float*>(                     // [G][TILE]
      smem_raw + (((size_t)G * Dk * sizeof(short) + 15) & ~(size_t)15));
  float* red = p_lds + (size_t)G * AD_BLOCK;                   // [max(G, NW)]

  for (int i = tid; i < G * Dk; i += AD_BLOCK) {
    int g = i / Dk, d = i % Dk;
    q_lds[i] = q[((long)b * Hq + h0 + g) * Dk + d];
  }
  __syncthreads();

  float m[G], l[G], acc[G * DVT];
#pragma unroll
  for (int g = 0; g < G; ++g) { m[g] = -1e30f; l[g] = 0.0f; }
#pragma unroll
  for (int i = 0; i < G * DVT; ++i) acc[i] = 0.0f;

  const long kbase = ((long)b * Hkv + hk) * Scap;
  int lo = (window > 0 && S > window) ? (S - window) : 0;
  int hi = S;
  if (nsplit > 1) {
    // static tile-aligned slices: capacity-based under graph capture
    // (S lives on device), S-based in eager mode (capacity-based
    // slicing would leave the tail slices empty)
    const long spanc = s_ptr ? Scap : (long)S;
    const long per = ((spanc + nsplit - 1) / nsplit + AD_BLOCK - 1)
                     / AD_BLOCK * AD_BLOCK;
    lo = max((long)lo, per * split);
    hi = min((long)S, per * (split + 1));
  }

  for (int tile = lo; tile < hi; tile += AD_BLOCK) {
    const int s_idx = tile + tid;
    float sc[G];
#pragma unroll
    for (int g = 0; g < G; ++g) sc[g] = -1e30f;
    if (s_idx < hi) {
      const short* krow = kcache + (kbase + s_idx) * Dk;
      float dot[G];
#pragma unroll
      for (int g = 0; g < G; ++g) dot[g] = 0.0f;
      // v_dot2c_f32_bf16: 2 bf16 products + f32 accumulate per VALU op
      // (q stays bf16 in LDS — multiplying bf16 inputs under f32
      // accumulation is numerically identical to the cvt+fma chain).
      // K is loaded in 64-element chunks BEFORE the LDS+dot block:
      // mixing a global load with LDS reads every 8 elements serializes
      // on a per-iteration vmcnt(0) (guide §5 trap 4(b)) — batching the
      // loads keeps 8 of them in flight per wait.
      int d0 = 0;
      for (; d0 + 64 <= Dk; d0 += 64) {
        bf16x8_t kc[8];
#pragma unroll
        for (int c = 0; c < 8; ++c)
          kc[c] = *reinterpret_cast<const bf16x8_t*>(krow + d0 + c * 8);
#pragma unroll
        for (int c = 0; c < 8; ++c)
#pragma unroll
          for (int g = 0; g < G; ++g) {
            bf16x8_t qv = *reinterpret_cast<const bf16x8_t*>(
                q_lds + (size_t)g * Dk + d0 + c * 8);
#pragma unroll
            for (int j = 0; j < 4; ++j)
              dot[g] = dot2_bf16(bf16x2_t{kc[c][2 * j], kc[c][2 * j + 1]},
                                 bf16x2_t{qv[2 * j], qv[2 * j + 1]}, dot[g]);
          }
      }
      for (; d0 < Dk; d0 += 8) {  // Dk % 64 tail (Dk % 8 == 0)
        bf16x8_t kv = *reinterpret_cast<const bf16x8_t*>(krow + d0);
#pragma unroll
        for (int g = 0; g < G; ++g) {
          bf16x8_t qv =
              *reinterpret_cast<const bf16x8_t*>(q_lds + (size_t)g * Dk + d0);
#pragma unroll
          for (int j = 0; j < 4; ++j)
            dot[g] = dot2_bf16(bf16x2_t{kv[2 * j], kv[2 * j + 1]},
                               bf16x2_t{qv[2 * j], qv[2 * j + 1]}, dot[g]);
        }
      }
#pragma unroll
      for (int g = 0; g < G; ++g) {
        float v = dot[g] * scale;
        if (softcap > 0.0f) v = softcap * tanhf(v / softcap);
        sc[g] = v;
      }
    }
    // Online-softmax reductions for ALL G heads in two staged passes
    // (4 barriers/tile): per-head block_max/block_sum loops cost 4*G
    // barriers and serialize the reductions — here wave w reduces
    // heads w, w+NW, ... in parallel from the staged p_lds rows.
    constexpr int NW = AD_BLOCK / WAVE;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    float alpha[G];
#pragma unroll
    for (int g = 0; g < G; ++g) p_lds[(size_t)g * AD_BLOCK + tid] = sc[g];
    __syncthreads();
    for (int g = wid; g < G; g += NW) {
      float tm = -1e30f;
#pragma unroll
      for (int i = 0; i < NW; ++i)
        tm = fmaxf(tm, p_lds[(size_t)g * AD_BLOCK + lane + i * WAVE]);
      tm = wave_max(tm);
      if (lane == 0) red[g] = tm;
    }
    __syncthreads();
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const float mnew = fmaxf(m[g], red[g]);
      alpha[g] = __expf(m[g] - mnew);
      float p = (s_idx < hi && sc[g] > -1e29f) ? __expf(sc[g] - mnew) : 0.0f;
      p_lds[(size_t)g * AD_BLOCK + tid] = p;
      m[g] = mnew;
    }
    __syncthreads();
    for (int g = wid; g < G; g += NW) {
      float ts = 0.0f;
#pragma unroll
      for (int i = 0; i < NW; ++i)
        ts += p_lds[(size_t)g * AD_BLOCK + lane + i * WAVE];
      ts = wave_sum(ts);
      if (lane == 0) red[g] = ts;
    }
    __syncthreads();
#pragma unroll
    for (int g = 0; g < G; ++g) l[g] = l[g] * alpha[g] + red[g];
    const int ntile = min(AD_BLOCK, hi - tile);
    const bf16* vbase = ((const bf16*)vcache) + (kbase + tile) * vstride + tid;
    float o[G * DVT];
#pragma unroll
    for (int i = 0; i < G * DVT; ++i) o[i] = 0.0f;
    if (tid < Dv) {
      // per-u column clamp + mask (loop-invariant: no per-element
      // guarded loads — guide §5 trap 4(c))
      long vcol[DVT];
      float vmask[DVT];
#pragma unroll
      for (int u = 0; u < DVT; ++u) {
        const int dv = tid + u * AD_BLOCK;
        vcol[u] = (dv < Dv) ? dv - tid : 0;
        vmask[u] = (dv < Dv) ? 1.0f : 0.0f;
      }
      // batch 8 keys of V loads per wait (trap 4(b), as in the K loop)
      int t = 0;
      for (; t + 8 <= ntile; t += 8) {
        float vv8[8][DVT];
#pragma unroll
        for (int tt = 0; tt < 8; ++tt)
#pragma unroll
          for (int u = 0; u < DVT; ++u)
            vv8[tt][u] =
                vmask[u] * bf2f(vbase[(long)(t + tt) * vstride + vcol[u]]);
#pragma unroll
        for (int tt = 0; tt < 8; ++tt)
#pragma unroll
          for (int g = 0; g < G; ++g) {
            const float p = p_lds[(size_t)g * AD_BLOCK + t + tt];
#pragma unroll
            for (int u = 0; u < DVT; ++u) o[g * DVT + u] += p * vv8[tt][u];
          }
      }
      for (; t < ntile; ++t) {
#pragma unroll
        for (int u = 0; u < DVT; ++u) {
          const float vv = vmask[u] * bf2f(vbase[(long)t * vstride + vcol[u]]);
#pragma unroll
          for (int g = 0; g < G; ++g)
            o[g * DVT + u] += p_lds[(size_t)g * AD_BLOCK + t] * vv;
        }
      }
    }
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
      for (int u = 0; u < DVT; ++u)
        acc[g * DVT + u] = acc[g * DVT + u] * alpha[g] + o[g * DVT + u];
    __syncthreads();
  }

  if (nsplit == 1) {
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
      for (int u = 0; u < DVT; ++u) {
        const int dv = tid + u * AD_BLOCK;
        if (dv < Dv)
          ((bf16*)out)[((long)b * Hq + h0 + g) * Dv + dv] =
              f2bf(acc[g * DVT + u] / l[g]);
      }
    return;
  }
  // partials: O unnormalized, plus (m, l)
#pragma unroll
  for (int g = 0; g < G; ++g)
#pragma unroll
    for (int u = 0; u < DVT; ++u) {
      const int dv = tid + u * AD_BLOCK;
      if (dv < Dv) {
        const long base = (((long)b * Hq + h0 + g) * nsplit + split) * Dv;
        part_o[base + dv] = acc[g * DVT + u];
      }
    }
  if (tid == 0) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const long base = (((long)b * Hq + h0 + g) * nsplit + split) * 2;
      part_ml[base + 0] = m[g];
      part_ml[base + 1] = l[g];
    }
  }
}

// Combine: one block per (b, hq); Dv threads reduce NS partials.
__global__ void attn_decode_combine_kernel(
    const float* __restrict__ part_o,  // [B*Hq, NS, Dv]
    const float* __restrict__ part_ml, // [B*Hq, NS, 2]
    short* __restrict__ out,           // [B, Hq, Dv]
    int NS, int Dv) {
  const long bh = blockIdx.x;
  const int tid = threadIdx.x;
  float mstar = -1e30f;
  for (int s = 0; s < NS; ++s)
    mstar = fmaxf(mstar, part_ml[(bh * NS + s) * 2]);
  float lsum = 0.0f;
  for (int s = 0; s < NS; ++s) {
    const float ms = part_ml[(bh * NS + s) * 2];
    const float ls = part_ml[(bh * NS + s) * 2 + 1];
    lsum += ls * __expf(ms - mstar);
  }
  for (int d = tid; d < Dv; d += blockDim.x) {
    float o = 0.0f;
    for (int s = 0; s < NS; ++s) {
      const float ms = part_ml[(bh * NS + s) * 2];
      o += part_o[(bh * NS + s) * Dv + d] * __expf(ms - mstar);
    }
    ((bf16*)out)[bh * Dv + d] = f2bf(o / lsum);
  }
}

extern "C" void launch_attn_decode(const void* q, const void* k, const void* v,
                                   void* out, float* part_o, float* part_ml,
                                   int nsplit, const int* s_ptr, int B, int Hq,
                                   int Hkv, int S, long Scap, int Dk, int Dv,
                                   long vstride, float scale, float softcap,
                                   int window, hipStream_t stream) {
  const int Gtot = Hq / Hkv;
  const int DVT = (Dv + AD_BLOCK - 1) / AD_BLOCK;
  // Big-G over few KV heads (absorbed MLA): split heads across
  // gridDim.z — smaller per-block register state, full-chip grids, K
  // tiles re-read from L2 by the sibling head-groups.  8-head groups
  // halve the K re-reads of the round-1 4-head choice at similar
  // occupancy (A/B'd on hardware).
  const int G = (Gtot == 16 && DVT == 2) ? 8 : Gtot;  // 16: -2% (A/B)
  const int NHG = Gtot / G;
  const int nred = (G > AD_BLOCK / WAVE) ? G : AD_BLOCK / WAVE;
  size_t smem = (((size_t)G * Dk * sizeof(short) + 15) & ~(size_t)15) +
                ((size_t)G * AD_BLOCK + nred) * sizeof(float);
  dim3 grid((unsigned)(B * Hkv), (unsigned)nsplit, (unsigned)NHG);
  dim3 block(AD_BLOCK);
#define AD_CASE(GG, VT)                                                      \
  case GG * 16 + VT:                                                         \
    attn_decode_kernel<GG, VT><<<grid, block, smem, stream>>>(               \
        (const short*)q, (const short*)k, (const short*)v, (short*)out,      \
        part_o, part_ml, s_ptr, B, Hq, Hkv, S, Scap, Dk, Dv, vstride, scale, \
        softcap, window);                                                    \
    break;
  switch (G * 16 + DVT) {
    AD_CASE(1, 1)
    AD_CASE(2, 1)
    AD_CASE(4, 1)
    AD_CASE(4, 2)  // absorbed MLA head-group variants, Dv = rank = 512
    AD_CASE(8, 2)
    AD_CASE(16, 2)
    AD_CASE(6, 1)
    AD_CASE(7, 1)  // qwen2 (28 q / 4 kv)
    AD_CASE(8, 1)
    AD_CASE(16, 1)
    default:
      break;
  }
#undef AD_CASE
  if (nsplit > 1) {
    attn_decode_combine_kernel<<<dim3((unsigned)(B * Hq)), dim3(128), 0,
                                 stream>>>(part_o, part_ml, (short*)out,
                                           nsplit, Dv);
  }
}

extern "C" bool attn_decode_supported_shape(int G, int Dv) {
  if (Dv > 512) return false;
  const int dvt = (Dv + AD_BLOCK - 1) / AD_BLOCK;
  if (dvt == 2) return G == 16 || G == 4;
  return G == 1 || G == 2 || G == 4 || G == 6 || G == 7 || G == 8 ||
         G == 16;
}
