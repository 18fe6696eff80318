// Flash-decode attention (T=1) over the KV cache — the decode-latency
// hot op (SURVEY.md §7 hard-part #2).
//
// Shapes: q [B, Hq, 1, Dk], kcache [B, Hkv, Scap, Dk], vcache
// [B, Hkv, Scap, Dv]; S = valid length.  GQA: one block serves one
// (batch, kv-head) pair and computes all G = Hq/Hkv query heads at
// once, so each K/V byte is read once regardless of the GQA ratio.
// Supports MLA shapes (Dk=192, Dv=128), gemma2 softcap + sliding
// window.  fp32 accumulation, online softmax over key tiles.
//
// G is a template parameter so all per-head register arrays stay
// statically indexed (guide §5.4 rule 20: runtime-indexed ext_vector
// arrays spill to scratch).
//
// Structure per tile of TILE=BLOCK keys: each thread owns one key row
// (its K row is a contiguous 2*Dk-byte read), computes G dot products
// against q (staged in LDS); block max/sum reduce; then the first Dv
// threads accumulate O[d] += sum_t p[t] * V[t][d] with per-key
// coalesced V row reads, p broadcast from LDS.

#include "hip_common.h"

#define AD_BLOCK 256

template <int G>
__global__ __launch_bounds__(AD_BLOCK) void attn_decode_kernel(
    const short* __restrict__ q,      // [B, Hq, Dk]
    const short* __restrict__ kcache, // [B, Hkv, Scap, Dk]
    const short* __restrict__ vcache, // [B, Hkv, Scap, Dv]
    short* __restrict__ out,          // [B, Hq, Dv]
    const int* __restrict__ s_ptr,    // device position (S = *s_ptr + 1), or null
    int B, int Hq, int Hkv, int S, long Scap, int Dk, int Dv, float scale,
    float softcap, int window) {
  if (s_ptr) S = *s_ptr + 1;  // hipGraph-captured decode: length lives on device
  const int b = blockIdx.x / Hkv;
  const int hk = blockIdx.x % Hkv;
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* q_lds = reinterpret_cast<float*>(smem_raw);           // [G][Dk]
  float* p_lds = q_lds + (size_t)G * Dk;                       // [G][TILE]
  float* red = p_lds + (size_t)G * AD_BLOCK;                   // [BLOCK/WAVE]

  for (int i = tid; i < G * Dk; i += AD_BLOCK) {
    int g = i / Dk, d = i % Dk;
    q_lds[i] = bf2f(((const bf16*)q)[((long)b * Hq + hk * G + g) * Dk + d]);
  }
  __syncthreads();

  float m[G], l[G], acc[G];
#pragma unroll
  for (int g = 0; g < G; ++g) { m[g] = -1e30f; l[g] = 0.0f; acc[g] = 0.0f; }

  const long kbase = ((long)b * Hkv + hk) * Scap;
  const int s_lo = (window > 0 && S > window) ? (S - window) : 0;

  for (int tile = s_lo; tile < S; tile += AD_BLOCK) {
    const int s_idx = tile + tid;
    // ---- scores: one key per thread, G dots ----
    float sc[G];
#pragma unroll
    for (int g = 0; g < G; ++g) sc[g] = -1e30f;
    if (s_idx < S) {
      const short* krow = kcache + (kbase + s_idx) * Dk;
      float dot[G];
#pragma unroll
      for (int g = 0; g < G; ++g) dot[g] = 0.0f;
      for (int d = 0; d < Dk; d += 4) {
        short4v kv = *reinterpret_cast<const short4v*>(krow + d);
        float k0 = bfbits2f(kv.x), k1 = bfbits2f(kv.y), k2 = bfbits2f(kv.z),
              k3 = bfbits2f(kv.w);
#pragma unroll
        for (int g = 0; g < G; ++g) {
          const float* qg = q_lds + (size_t)g * Dk + d;
          dot[g] += k0 * qg[0] + k1 * qg[1] + k2 * qg[2] + k3 * qg[3];
        }
      }
#pragma unroll
      for (int g = 0; g < G; ++g) {
        float v = dot[g] * scale;
        if (softcap > 0.0f) v = softcap * tanhf(v / softcap);
        sc[g] = v;
      }
    }
    // ---- online softmax per head ----
    float alpha[G];
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float tmax = block_max<AD_BLOCK>(sc[g], red);
      float mnew = fmaxf(m[g], tmax);
      alpha[g] = __expf(m[g] - mnew);
      float p = (s_idx < S) ? __expf(sc[g] - mnew) : 0.0f;
      p_lds[(size_t)g * AD_BLOCK + tid] = p;
      float psum = block_sum<AD_BLOCK>(p, red);
      l[g] = l[g] * alpha[g] + psum;
      m[g] = mnew;
    }
    __syncthreads();
    // ---- O update ----
    const int ntile = min(AD_BLOCK, S - tile);
    if (tid < Dv) {
      const bf16* vbase = ((const bf16*)vcache) + (kbase + tile) * Dv + tid;
      float o[G];
#pragma unroll
      for (int g = 0; g < G; ++g) o[g] = 0.0f;
      for (int t = 0; t < ntile; ++t) {
        float vv = bf2f(vbase[(long)t * Dv]);
#pragma unroll
        for (int g = 0; g < G; ++g) o[g] += p_lds[(size_t)g * AD_BLOCK + t] * vv;
      }
#pragma unroll
      for (int g = 0; g < G; ++g) acc[g] = acc[g] * alpha[g] + o[g];
    } else {
#pragma unroll
      for (int g = 0; g < G; ++g) acc[g] *= alpha[g];
    }
    __syncthreads();
  }

  if (tid < Dv) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float o = acc[g] / l[g];
      ((bf16*)out)[((long)b * Hq + hk * G + g) * Dv + tid] = f2bf(o);
    }
  }
}

extern "C" void launch_attn_decode(const void* q, const void* k, const void* v,
                                   void* out, const int* s_ptr, int B, int Hq,
                                   int Hkv, int S, long Scap, int Dk, int Dv,
                                   float scale, float softcap, int window,
                                   hipStream_t stream) {
  const int G = Hq / Hkv;
  size_t smem = ((size_t)G * Dk + (size_t)G * AD_BLOCK + AD_BLOCK / WAVE) *
                sizeof(float);
  dim3 grid((unsigned)(B * Hkv));
  dim3 block(AD_BLOCK);
#define AD_CASE(GG)                                                          \
  case GG:                                                                   \
    attn_decode_kernel<GG><<<grid, block, smem, stream>>>(                   \
        (const short*)q, (const short*)k, (const short*)v, (short*)out,      \
        s_ptr, B, Hq, Hkv, S, Scap, Dk, Dv, scale, softcap, window);         \
    break;
  switch (G) {
    AD_CASE(1)
    AD_CASE(2)
    AD_CASE(4)
    AD_CASE(6)
    AD_CASE(8)
    AD_CASE(16)
    default:
      // unsupported GQA ratio — caller checks and falls back loudly
      break;
  }
#undef AD_CASE
}

extern "C" bool attn_decode_supported_ratio(int G) {
  return G == 1 || G == 2 || G == 4 || G == 6 || G == 8 || G == 16;
}
