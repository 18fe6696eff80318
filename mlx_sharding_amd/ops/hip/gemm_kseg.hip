// K-segmented M<=64 dense bf16 GEMM for K-long decode projections.
//
// Target: the llama-style down_proj shapes ([8192 out, 28672 k],
// [4096, 14336]) where hipBLASLt plateaus at ~3.3 TB/s while reaching
// ~5.1 on the transposed-aspect shapes (tools/gemm_layout_probe.py).
//
// Why previous in-house dense GEMMs lost (docs/PERFORMANCE.md):
// loading A straight in MFMA fragment layout makes every wave
// instruction touch 16 weight rows x 16 B — at llama scale that is
// ~100k concurrent fine-grained row streams and DRAM page locality
// collapses (measured 1.0-1.5 TB/s).  This kernel changes exactly one
// thing: the A (weight) stream is COALESCED — whole-block cooperative
// row staging, 512 B-per-row runs into LDS, and MFMA fragments are
// read from LDS.  B (activations, <=64 rows) stays direct-from-global
// in fragment layout: x is L2-resident and that pattern is proven in
// the MoE kernels (moe.hip).
//
// C[M<=64, N] = X[M, K] @ W[N, K]^T, fp32 atomic accumulation over
// grid.z K-segments, bf16 convert by a follow-up elementwise kernel.

#include <hip/hip_runtime.h>

#define GK_WAVE 64
#define GK_WAVES 4
#define GK_BLOCK (GK_WAVE * GK_WAVES)
#define GK_KC 128                 // k elems staged per chunk
#define GK_STAGE (GK_KC / 32)     // 16B-units per thread per chunk
#define GK_PAD 8                  // LDS row pad (elems): stride 264 -> 4-bank row shift
#define GK_LDS_STRIDE (GK_KC + GK_PAD)

typedef __bf16 gkbf16x8 __attribute__((ext_vector_type(8)));
typedef float gkf32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ unsigned short gk_f2bf(float f) {
  union { float f; unsigned int u; } v{f};
  unsigned int lsb = (v.u >> 16) & 1;
  return (unsigned short)((v.u + 0x7fff + lsb) >> 16);
}

// grid = (ceil(N/64), 1, ksegs); block = 256.
// kseg_len % GK_KC == 0 (host guarantees; last segment may be short but
// still KC-aligned because K % GK_KC == 0).
__global__ __launch_bounds__(GK_BLOCK) void gemm_kseg_kernel(
    const short* __restrict__ x,   // [M, K] bf16
    const short* __restrict__ w,   // [N, K] bf16
    float* __restrict__ out,       // [64, N] fp32 (zeroed when ksegs > 1)
    int M, int N, int K, int kseg_len, int ksegs) {
  const int n0 = blockIdx.x * 64;
  const int k0 = blockIdx.z * kseg_len;
  const int k1 = min(k0 + kseg_len, K);
  if (n0 >= N || k0 >= k1) return;
  const int tid = threadIdx.x;
  const int lane = tid & (GK_WAVE - 1);
  const int wid = tid / GK_WAVE;

  __shared__ short a_lds[64 * GK_LDS_STRIDE];
  __shared__ short b_lds[64 * GK_LDS_STRIDE];

  // accumulators: 4 token-tiles x f32x4 (wave owns rows 16*wid..+15)
  gkf32x4 acc0 = {0, 0, 0, 0}, acc1 = {0, 0, 0, 0};
  gkf32x4 acc2 = {0, 0, 0, 0}, acc3 = {0, 0, 0, 0};

  const int r_base = n0 + 16 * wid;     // this wave's first A row
  const int a_row_frag = 16 * wid + (lane & 15);  // LDS row for A frags
  const int kq = (lane >> 4) * 8;       // k-offset within a 32-k block

  // staging assignment: unit u covers row u/32, bytes (u%32)*16 of the
  // chunk — 32 consecutive threads write one row's 512 B run
  // (coalesced 1 KB per wave instruction).  Register-prefetch
  // pipeline: chunk i+1's 8 global loads are issued while chunk i
  // computes, hiding DRAM latency behind the MFMA block without a
  // second LDS buffer (barrier count unchanged, occupancy stays 4
  // workgroups/CU).
  int stage_row[GK_STAGE], stage_sw[GK_STAGE];
  long stage_goff[GK_STAGE], stage_xoff[GK_STAGE];
#pragma unroll
  for (int i = 0; i < GK_STAGE; ++i) {
    const int u = i * GK_BLOCK + tid;
    const int row = u / (GK_KC / 8);
    const int unit = u % (GK_KC / 8);
    stage_row[i] = row;
    stage_sw[i] = unit * 8;  // plain placement: an XOR swizzle measured
    //                            5x MORE conflict cycles (PMC), reverted
    stage_goff[i] = (long)min(n0 + row, N - 1) * K + unit * 8;
    stage_xoff[i] = (long)min(row, M - 1) * K + unit * 8;
  }
  gkbf16x8 pre[GK_STAGE], pre_b[GK_STAGE];
#pragma unroll
  for (int i = 0; i < GK_STAGE; ++i) {
    pre[i] = *reinterpret_cast<const gkbf16x8*>(w + stage_goff[i] + k0);
    pre_b[i] = *reinterpret_cast<const gkbf16x8*>(x + stage_xoff[i] + k0);
  }

  for (int kc = k0; kc < k1; kc += GK_KC) {
#pragma unroll
    for (int i = 0; i < GK_STAGE; ++i) {
      *reinterpret_cast<gkbf16x8*>(
          a_lds + stage_row[i] * GK_LDS_STRIDE + stage_sw[i]) = pre[i];
      *reinterpret_cast<gkbf16x8*>(
          b_lds + stage_row[i] * GK_LDS_STRIDE + stage_sw[i]) = pre_b[i];
    }
    __syncthreads();
    if (kc + GK_KC < k1) {
#pragma unroll
      for (int i = 0; i < GK_STAGE; ++i) {
        pre[i] = *reinterpret_cast<const gkbf16x8*>(
            w + stage_goff[i] + kc + GK_KC);
        pre_b[i] = *reinterpret_cast<const gkbf16x8*>(
            x + stage_xoff[i] + kc + GK_KC);
      }
    }

    // ---- MFMA over the chunk ---------------------------------------
#pragma unroll
    for (int kb = 0; kb < GK_KC / 32; ++kb) {
      const int ko = kb * 32 + kq;
      const gkbf16x8 a = *reinterpret_cast<const gkbf16x8*>(
          a_lds + a_row_frag * GK_LDS_STRIDE + ko);
      const int tr = lane & 15;
      const gkbf16x8 b0 = *reinterpret_cast<const gkbf16x8*>(
          b_lds + (0 + tr) * GK_LDS_STRIDE + ko);
      const gkbf16x8 b1 = *reinterpret_cast<const gkbf16x8*>(
          b_lds + (16 + tr) * GK_LDS_STRIDE + ko);
      const gkbf16x8 b2 = *reinterpret_cast<const gkbf16x8*>(
          b_lds + (32 + tr) * GK_LDS_STRIDE + ko);
      const gkbf16x8 b3 = *reinterpret_cast<const gkbf16x8*>(
          b_lds + (48 + tr) * GK_LDS_STRIDE + ko);
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b0, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b1, acc1, 0, 0, 0);
      acc2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b2, acc2, 0, 0, 0);
      acc3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b3, acc3, 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: C row = r_base + (lane>>4)*4 + reg, token = tile*16
  // + (lane&15) (same layout convention as moe.hip's MFMA epilogues)
  const gkf32x4 accs[4] = {acc0, acc1, acc2, acc3};
#pragma unroll
  for (int tt = 0; tt < 4; ++tt) {
    const int tok = tt * 16 + (lane & 15);
    if (tok >= M) continue;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int o = r_base + (lane >> 4) * 4 + reg;
      if (o < N) {
        if (ksegs > 1)
          atomicAdd(&out[(long)tok * N + o], accs[tt][reg]);
        else
          out[(long)tok * N + o] = accs[tt][reg];
      }
    }
  }
}

__global__ void gemm_kseg_zero(float* __restrict__ p, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = 0.0f;
}

__global__ void gemm_kseg_f32_to_bf16(const float* __restrict__ in,
                                      short* __restrict__ o, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) o[i] = (short)gk_f2bf(in[i]);
}

extern "C" void launch_gemm_kseg(const void* x, const void* w, void* out_f32,
                                 void* out_bf16, int M, int N, int K,
                                 int ksegs, hipStream_t stream) {
  const int kseg_len = ((K / ksegs + GK_KC - 1) / GK_KC) * GK_KC;
  ksegs = (K + kseg_len - 1) / kseg_len;  // actual segments after rounding
  if (ksegs > 1) {
    // zero via a kernel, NOT hipMemsetAsync: a memset enqueued during
    // hipGraph stream capture is not recorded into the graph, so
    // replays would accumulate into stale partials (caught by
    // test_gemm_kseg_under_graph_capture)
    const long zn = (long)64 * N;
    gemm_kseg_zero<<<dim3((unsigned)((zn + 255) / 256)), dim3(256), 0,
                     stream>>>((float*)out_f32, zn);
  }
  dim3 grid((unsigned)((N + 63) / 64), 1, (unsigned)ksegs);
  gemm_kseg_kernel<<<grid, dim3(GK_BLOCK), 0, stream>>>(
      (const short*)x, (const short*)w, (float*)out_f32, M, N, K, kseg_len,
      ksegs);
  const long n = (long)M * N;
  gemm_kseg_f32_to_bf16<<<dim3((unsigned)((n + 255) / 256)), dim3(256), 0,
                          stream>>>((const float*)out_f32, (short*)out_bf16,
                                    n);
}

// ---------------------------------------------------------------------------
// w4a16 variant: packed int4 weights (4x less DRAM than the bf16
// kernel) with the fp16 magic-number dequant from moe_w4f16.hip,
// dequantized at FRAGMENT-read time (after the coalesced LDS round
// trip of the packed words).  Activations arrive fp16; MFMA is
// v_mfma_f32_16x16x32_f16.  Makes huge-K int4 projections (llama-70B
// class) stream at packed-weight bandwidth — the LDS-free w4f16_gemv
// collapses there (docs/PERFORMANCE.md).
// ---------------------------------------------------------------------------

#include "hip_common.h"

typedef _Float16 gk16x2 __attribute__((ext_vector_type(2)));
typedef _Float16 gk16x8 __attribute__((ext_vector_type(8)));

union gk_f16pack { unsigned int u; gk16x2 h; };

// dequant one repacked u32 (8 x 4-bit, natural k-order) — same scheme
// as moe_w4f16.hip:dq8 (OR 0x6400 -> exact 1024+q; pk_add -1032 ->
// exact q-8; pk_fma applies (s, b+8s))
__device__ __forceinline__ void gk_dq8(unsigned int w, gk16x2 s2, gk16x2 b2,
                                       gk16x8* out) {
  gk16x2* op = reinterpret_cast<gk16x2*>(out);
  const gk16x2 c = {(_Float16)(-1032.0f), (_Float16)(-1032.0f)};
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    gk_f16pack p;
    p.u = ((w >> (4 * j)) & 0x000F000Fu) | 0x64006400u;
    op[j] = (p.h + c) * s2 + b2;
  }
}

__device__ __forceinline__ gk16x2 gk_splat2(float v) {
  const _Float16 h = (_Float16)v;
  return (gk16x2){h, h};
}

#define GKW_KC 256                       // k elems per staged chunk
#define GKW_AW (GKW_KC / 8)              // packed words per row per chunk
#define GKW_AS (GKW_AW + 2)              // padded LDS row stride (words)
#define GKW_BS (GKW_KC + 8)              // fp16 B row stride (elems)

template <int GS>
__global__ __launch_bounds__(GK_BLOCK) void gemm_kseg_w4_kernel(
    const _Float16* __restrict__ x,       // [M, K] fp16
    const unsigned int* __restrict__ wq,  // [N, K/8] repacked
    const short* __restrict__ sc,         // [N, K/GS] bf16 bits
    const short* __restrict__ bi,         // [N, K/GS]
    float* __restrict__ out,              // [64, N] fp32
    int M, int N, int K, int kseg_len, int ksegs) {
  const int n0 = blockIdx.x * 64;
  const int k0 = blockIdx.z * kseg_len;
  const int k1 = min(k0 + kseg_len, K);
  if (n0 >= N || k0 >= k1) return;
  const int tid = threadIdx.x;
  const int lane = tid & (GK_WAVE - 1);
  const int wid = tid / GK_WAVE;
  const int wpr = K / 8;                  // packed words per full row
  const int ngr = K / GS;

  __shared__ unsigned int a_lds[64 * GKW_AS];
  __shared__ _Float16 b_lds[64 * GKW_BS];

  gkf32x4 acc0 = {0, 0, 0, 0}, acc1 = {0, 0, 0, 0};
  gkf32x4 acc2 = {0, 0, 0, 0}, acc3 = {0, 0, 0, 0};

  const int r_base = n0 + 16 * wid;
  const int a_row_frag = 16 * wid + (lane & 15);
  const int a_grow = min(n0 + a_row_frag, N - 1);  // global row for scales
  const int kq = (lane >> 4) * 8;

  // A staging: 64 rows x GKW_AW words = 16 B-units: 64*(GKW_AW/4)
  // units / 256 threads
  constexpr int AST = 64 * (GKW_AW / 4) / GK_BLOCK;  // units per thread
  int a_row[AST], a_unit[AST];
  long a_goff[AST];
#pragma unroll
  for (int i = 0; i < AST; ++i) {
    const int u = i * GK_BLOCK + tid;
    a_row[i] = u / (GKW_AW / 4);
    a_unit[i] = u % (GKW_AW / 4);
    a_goff[i] = (long)min(n0 + a_row[i], N - 1) * wpr + a_unit[i] * 4;
  }
  // B staging: 64 rows x KC elems fp16 = 64*KC*2/16 units
  constexpr int BST = 64 * (GKW_KC / 8) / GK_BLOCK;
  int b_row[BST], b_unit[BST];
  long b_goff[BST];
#pragma unroll
  for (int i = 0; i < BST; ++i) {
    const int u = i * GK_BLOCK + tid;
    b_row[i] = u / (GKW_KC / 8);
    b_unit[i] = u % (GKW_KC / 8);
    b_goff[i] = (long)min(b_row[i], M - 1) * K + b_unit[i] * 8;
  }

  uint4 pre_a[AST];
  gk16x8 pre_b[BST];
#pragma unroll
  for (int i = 0; i < AST; ++i)
    pre_a[i] = *reinterpret_cast<const uint4*>(wq + a_goff[i] + k0 / 8);
#pragma unroll
  for (int i = 0; i < BST; ++i)
    pre_b[i] = *reinterpret_cast<const gk16x8*>(x + b_goff[i] + k0);

  for (int kc = k0; kc < k1; kc += GKW_KC) {
#pragma unroll
    for (int i = 0; i < AST; ++i)
      *reinterpret_cast<uint4*>(a_lds + a_row[i] * GKW_AS + a_unit[i] * 4) =
          pre_a[i];
#pragma unroll
    for (int i = 0; i < BST; ++i)
      *reinterpret_cast<gk16x8*>(b_lds + b_row[i] * GKW_BS + b_unit[i] * 8) =
          pre_b[i];
    __syncthreads();
    // hoist this chunk's scale/bias pairs out of the MFMA loop: the
    // lane's rows touch ceil(KC/GS) (+1 alignment) groups per chunk;
    // loading them per-kb put an L2-latency dependency in front of
    // every dequant (measured 0.8 TB/s vs 4 expected)
    constexpr int NG = GKW_KC / GS + 1;
    gk16x2 cs2[NG], cb2[NG];
    {
      const int gbase = kc / GS;
#pragma unroll
      for (int c = 0; c < NG; ++c) {
        const int g = min(gbase + c, ngr - 1);
        const float sf = bfbits2f(sc[(long)a_grow * ngr + g]);
        cs2[c] = gk_splat2(sf);
        cb2[c] = gk_splat2(bfbits2f(bi[(long)a_grow * ngr + g]) + 8.0f * sf);
      }
    }
    if (kc + GKW_KC < k1) {
#pragma unroll
      for (int i = 0; i < AST; ++i)
        pre_a[i] = *reinterpret_cast<const uint4*>(
            wq + a_goff[i] + (kc + GKW_KC) / 8);
#pragma unroll
      for (int i = 0; i < BST; ++i)
        pre_b[i] = *reinterpret_cast<const gk16x8*>(
            x + b_goff[i] + kc + GKW_KC);
    }

#pragma unroll
    for (int kb = 0; kb < GKW_KC / 32; ++kb) {
      const int ko = kb * 32 + kq;          // k offset in chunk
      const int gi = ko / GS;               // group index within chunk
      const gk16x2 s2 = cs2[gi];
      const gk16x2 b2 = cb2[gi];
      const unsigned int aw =
          a_lds[a_row_frag * GKW_AS + ko / 8];
      gk16x8 a;
      gk_dq8(aw, s2, b2, &a);
      const int tr = lane & 15;
      const gk16x8 b0 = *reinterpret_cast<const gk16x8*>(
          b_lds + (0 + tr) * GKW_BS + ko);
      const gk16x8 b1 = *reinterpret_cast<const gk16x8*>(
          b_lds + (16 + tr) * GKW_BS + ko);
      const gk16x8 b2f = *reinterpret_cast<const gk16x8*>(
          b_lds + (32 + tr) * GKW_BS + ko);
      const gk16x8 b3 = *reinterpret_cast<const gk16x8*>(
          b_lds + (48 + tr) * GKW_BS + ko);
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b0, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b1, acc1, 0, 0, 0);
      acc2 = __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b2f, acc2, 0, 0, 0);
      acc3 = __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b3, acc3, 0, 0, 0);
    }
    __syncthreads();
  }

  const gkf32x4 accs[4] = {acc0, acc1, acc2, acc3};
#pragma unroll
  for (int tt = 0; tt < 4; ++tt) {
    const int tok = tt * 16 + (lane & 15);
    if (tok >= M) continue;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int o = r_base + (lane >> 4) * 4 + reg;
      if (o < N) {
        if (ksegs > 1)
          atomicAdd(&out[(long)tok * N + o], accs[tt][reg]);
        else
          out[(long)tok * N + o] = accs[tt][reg];
      }
    }
  }
}

extern "C" void launch_gemm_kseg_w4(const void* x, const void* wq,
                                    const void* sc, const void* bi,
                                    void* out_f32, void* out_bf16, int M,
                                    int N, int K, int gs, int ksegs,
                                    hipStream_t stream) {
  const int kseg_len = ((K / ksegs + GKW_KC - 1) / GKW_KC) * GKW_KC;
  ksegs = (K + kseg_len - 1) / kseg_len;
  if (ksegs > 1) {
    const long zn = (long)64 * N;
    gemm_kseg_zero<<<dim3((unsigned)((zn + 255) / 256)), dim3(256), 0,
                     stream>>>((float*)out_f32, zn);
  }
  dim3 grid((unsigned)((N + 63) / 64), 1, (unsigned)ksegs);
#define GKW_CASE(GSV)                                                        \
  gemm_kseg_w4_kernel<GSV><<<grid, dim3(GK_BLOCK), 0, stream>>>(             \
      (const _Float16*)x, (const unsigned int*)wq, (const short*)sc,         \
      (const short*)bi, (float*)out_f32, M, N, K, kseg_len, ksegs)
  if (gs == 32) GKW_CASE(32);
  else if (gs == 64) GKW_CASE(64);
  else GKW_CASE(128);
#undef GKW_CASE
  const long n = (long)M * N;
  gemm_kseg_f32_to_bf16<<<dim3((unsigned)((n + 255) / 256)), dim3(256), 0,
                          stream>>>((const float*)out_f32, (short*)out_bf16,
                                    n);
}
