// Flash-style prefill attention on MFMA (gfx950 mfma_f32_16x16x32_bf16).
//
// Never materializes the S×S score matrix (the reference's O(T²) mask +
// full-logit path): per 64-row Q tile, iterate 32-key
// K/V tiles computing QK^T → online softmax → P·V with fp32 running
// (m, l) per row — guide Appendix B "fused attention prefill".
//
// Fragment maps (verified by the mfma_probe binding + numerics tests):
//   mfma_f32_16x16x32_bf16: A(16x32) lane: row=l&15, k=(l>>4)*8+j
//                           B(32x16) lane: col=l&15, k=(l>>4)*8+j
//                           C/D:      lane: col=l&15, row=(l>>4)*4+reg
// Both A and B operands load as ONE contiguous 16-byte read per lane
// from row-major [rows][Dk] tensors (B is K^T with K row-major — same
// pattern).  P (in C layout) is redistributed to A layout through a
// small LDS tile per wave.
//
// Block = 4 waves; each wave owns 16 q rows; K/V tiles are 32 keys.
// Causal offset, GQA (q head -> kv head), softcap and sliding window
// all supported; out-of-range q rows masked.

#include "hip_common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define AP_WAVES 4
#define AP_BLOCK (AP_WAVES * WAVE)
#define AP_QTILE (AP_WAVES * 16)  // 64 q rows per block
#define AP_KTILE 32
#define AP_MAXKS 8   // Dk <= 256
#define AP_MAXDH 16  // Dv <= 256

template <int NKS, int NDH>
__global__ __launch_bounds__(AP_BLOCK) void attn_prefill_kernel(
    const short* __restrict__ q,   // [B, Hq, T, Dk]
    const short* __restrict__ k,   // [B, Hkv, S, Dk] (row-contiguous)
    const short* __restrict__ v,   // [B, Hkv, S, Dv]
    short* __restrict__ out,       // [B, Hq, T, Dv]
    int B, int Hq, int Hkv, int T, int S, long kScap, long vScap, int Dk,
    int Dv, float scale, float softcap, int window, int causal_offset) {
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int h = bh % Hq;
  const int hk = h / (Hq / Hkv);
  const int q0 = blockIdx.x * AP_QTILE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int wq0 = q0 + wid * 16;       // this wave's first q row

  // (out-of-range waves still participate in barriers/V staging)

  const short* qbase = q + (((long)b * Hq + h) * T) * Dk;
  const short* kbase = k + ((long)b * Hkv + hk) * kScap * Dk;
  const short* vbase = v + ((long)b * Hkv + hk) * vScap * Dv;

  // LDS: per-wave P tile [16][32] bf16 + block-shared TRANSPOSED V tile
  // [Dv][AP_KTILE+8] — the PV B-fragment needs V[k, col] for 8
  // consecutive k, which in key-major layout was 8 scalar LDS reads per
  // dh per lane (64/lane/tile); value-major rows make it ONE b128 read.
  // +8 row padding staggers the bank mapping of the 16 rows a quarter-
  // wave group reads.
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  short* p_lds = reinterpret_cast<short*>(smem_raw) + wid * 16 * AP_KTILE;
  short* v_lds = reinterpret_cast<short*>(smem_raw) + AP_WAVES * 16 * AP_KTILE;
  constexpr int VTS = AP_KTILE + 8;  // transposed-row stride (16B-aligned)

  // ---- load Q fragments (A layout): lane: row wq0+(l&15), 16B at kslice ----
  bf16x8 qfrag[NKS];
  {
    int row = min(wq0 + (lane & 15), T - 1);  // clamped; masked on write
    const short* qr = qbase + (long)row * Dk + (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < NKS; ++ks)
      qfrag[ks] = *reinterpret_cast<const bf16x8*>(qr + ks * 32);
  }

  float m_run[4], l_run[4];
  f32x4 oacc[NDH];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.0f; }
#pragma unroll
  for (int dh = 0; dh < NDH; ++dh) oacc[dh] = f32x4{0, 0, 0, 0};

  // causal bounds: per-wave compute range, block-uniform loop (the V
  // tile is staged once per block and shared by all four waves)
  const int wave_s_hi = min(S, causal_offset + min(wq0 + 15, T - 1) + 1);
  const int blk_s_hi = min(S, causal_offset + min(q0 + AP_QTILE - 1, T - 1) + 1);
  int s_lo = 0;
  if (window > 0)
    s_lo = max(0, causal_offset + q0 - window + 1) & ~(AP_KTILE - 1);

  for (int t0 = s_lo; t0 < blk_s_hi; t0 += AP_KTILE) {
    // ---- stage V tile transposed [Dv][KTILE] (coalesced short4 global
    // reads, scalar LDS scatter writes — 8 KB total) ----
    __syncthreads();
    {
      const int n4 = AP_KTILE * Dv / 4;
      for (int i = threadIdx.x; i < n4; i += AP_BLOCK) {
        const int kk = (i * 4) / Dv;
        const int key = t0 + kk;
        const int d = (i * 4) % Dv;
        short4v val =
            (key < S)
                ? *reinterpret_cast<const short4v*>(vbase + (long)key * Dv + d)
                : short4v{0, 0, 0, 0};
        v_lds[(d + 0) * VTS + kk] = val.x;
        v_lds[(d + 1) * VTS + kk] = val.y;
        v_lds[(d + 2) * VTS + kk] = val.z;
        v_lds[(d + 3) * VTS + kk] = val.w;
      }
    }
    __syncthreads();
    if (t0 >= wave_s_hi) continue;  // this wave's rows see no keys here
    // ---- QK^T: two 16-key column halves ----
    f32x4 c0 = {0, 0, 0, 0}, c1 = {0, 0, 0, 0};
    {
      const int key0 = t0 + (lane & 15);
      const int key1 = t0 + 16 + (lane & 15);
      const short* kr0 = kbase + (long)min(key0, S - 1) * Dk + (lane >> 4) * 8;
      const short* kr1 = kbase + (long)min(key1, S - 1) * Dk + (lane >> 4) * 8;
#pragma unroll
      for (int ks = 0; ks < NKS; ++ks) {
        bf16x8 kf0 = *reinterpret_cast<const bf16x8*>(kr0 + ks * 32);
        bf16x8 kf1 = *reinterpret_cast<const bf16x8*>(kr1 + ks * 32);
        c0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], kf0, c0, 0, 0, 0);
        c1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], kf1, c1, 0, 0, 0);
      }
    }
    // ---- scale, softcap, mask (element: col=l&15(+16), row=(l>>4)*4+reg) ----
    const int col = lane & 15;
    float sc[8];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) { sc[reg] = c0[reg]; sc[4 + reg] = c1[reg]; }
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int key = t0 + half * 16 + col;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = wq0 + (lane >> 4) * 4 + reg;
        float sv = sc[half * 4 + reg] * scale;
        if (softcap > 0.0f) sv = softcap * tanhf(sv / softcap);
        const int qpos = causal_offset + row;
        bool dead = (key >= S) || (key > qpos) || (row >= T);
        if (window > 0 && key <= qpos - window) dead = true;
        sc[half * 4 + reg] = dead ? -1e30f : sv;
      }
    }
    // ---- per-row online softmax (rows live in 16-lane groups) ----
    float alpha[4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      float mx = fmaxf(sc[reg], sc[4 + reg]);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
      float mnew = fmaxf(m_run[reg], mx);
      alpha[reg] = __expf(m_run[reg] - mnew);
      m_run[reg] = mnew;
      // guard: a row whose keys are ALL dead so far has mnew = -1e30 and
      // exp(-1e30 - -1e30) = 1 — dead scores must stay exactly 0.
      float p0 = (sc[reg] <= -1e29f) ? 0.0f : __expf(sc[reg] - mnew);
      float p1 = (sc[4 + reg] <= -1e29f) ? 0.0f : __expf(sc[4 + reg] - mnew);
      sc[reg] = p0;
      sc[4 + reg] = p1;
      float ps = p0 + p1;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        ps += __shfl_xor(ps, off, WAVE);
      l_run[reg] = l_run[reg] * alpha[reg] + ps;
    }
    // ---- stage P to LDS (C layout -> A layout) ----
#pragma unroll
    for (int half = 0; half < 2; ++half)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = (lane >> 4) * 4 + reg;
        p_lds[row * AP_KTILE + half * 16 + col] =
            (short)__bfloat16_as_ushort(f2bf(sc[half * 4 + reg]));
      }
    // (same-wave LDS write->read: hipcc inserts the lgkmcnt wait itself)
    // ---- P @ V ----
    bf16x8 pfrag = *reinterpret_cast<const bf16x8*>(
        p_lds + (lane & 15) * AP_KTILE + (lane >> 4) * 8);
#pragma unroll
    for (int dh = 0; dh < NDH; ++dh) {
      // B-frag of V: one contiguous b128 read from the transposed row
      bf16x8 vf = *reinterpret_cast<const bf16x8*>(
          v_lds + (dh * 16 + (lane & 15)) * VTS + (lane >> 4) * 8);
      f32x4 prev = oacc[dh];
      f32x4 scaled;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) scaled[reg] = prev[reg] * alpha[reg];
      oacc[dh] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vf, scaled, 0, 0, 0);
    }
  }

  // ---- epilogue: O /= l, write (col=l&15 -> d, row per reg) ----
  const int col = lane & 15;
  float rinv[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg)
    rinv[reg] = __builtin_amdgcn_rcpf(l_run[reg]);  // ~1ulp: fine for bf16 out
#pragma unroll
  for (int dh = 0; dh < NDH; ++dh) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row = wq0 + (lane >> 4) * 4 + reg;
      if (row < T) {
        float val = oacc[dh][reg] * rinv[reg];
        out[(((long)b * Hq + h) * T + row) * Dv + dh * 16 + col] =
            (short)__bfloat16_as_ushort(f2bf(val));
      }
    }
  }
}

// The launcher dispatches only instantiated (Dk/32, Dv/16) combos; a
// silent fall-through once returned an UNINITIALIZED output tensor
// (debug-llama Dk=Dv=32) — callers must check this first.
extern "C" bool attn_prefill_supported(int Dk, int Dv) {
  if (Dk % 32 != 0 || Dv % 16 != 0 || Dk > 256 || Dv > 256) return false;
  const int nks = Dk / 32, ndh = Dv / 16;
  switch (nks * 100 + ndh) {
    case 101: case 102: case 104:
    case 202: case 204: case 208:
    case 402: case 404: case 408:
    case 604: case 608:
    case 816:
      return true;
    default:
      return false;
  }
}

extern "C" void launch_attn_prefill(const void* q, const void* k, const void* v,
                                    void* out, int B, int Hq, int Hkv, int T,
                                    int S, long kScap, long vScap, int Dk,
                                    int Dv, float scale, float softcap,
                                    int window, int causal_offset,
                                    hipStream_t stream) {
  dim3 grid((T + AP_QTILE - 1) / AP_QTILE, B * Hq);
  size_t smem = (AP_WAVES * 16 * AP_KTILE + (AP_KTILE + 8) * Dv) * sizeof(short);
#define AP_CASE(KS, DH)                                                       \
  attn_prefill_kernel<KS, DH><<<grid, dim3(AP_BLOCK), smem, stream>>>(        \
      (const short*)q, (const short*)k, (const short*)v, (short*)out, B, Hq,  \
      Hkv, T, S, kScap, vScap, Dk, Dv, scale, softcap, window, causal_offset)
  const int nks = Dk / 32, ndh = Dv / 16;
  if (nks == 1 && ndh == 1) AP_CASE(1, 1);
  else if (nks == 1 && ndh == 2) AP_CASE(1, 2);
  else if (nks == 1 && ndh == 4) AP_CASE(1, 4);
  else if (nks == 2 && ndh == 2) AP_CASE(2, 2);
  else if (nks == 2 && ndh == 4) AP_CASE(2, 4);
  else if (nks == 2 && ndh == 8) AP_CASE(2, 8);
  else if (nks == 4 && ndh == 2) AP_CASE(4, 2);
  else if (nks == 4 && ndh == 4) AP_CASE(4, 4);
  else if (nks == 4 && ndh == 8) AP_CASE(4, 8);
  else if (nks == 6 && ndh == 8) AP_CASE(6, 8);
  else if (nks == 6 && ndh == 4) AP_CASE(6, 4);
  else if (nks == 8 && ndh == 16) AP_CASE(8, 16);  // gemma2 256/256 heads
#undef AP_CASE
}

// ---------------------------------------------------------------------------
// Fragment-layout probe: D = A @ B for one 16x32 @ 32x16 mfma, using the
// maps documented above.  The numerics test compares against torch.matmul
// with asymmetric operands (guide G9: transpose-detecting).
// ---------------------------------------------------------------------------
__global__ void mfma_probe_kernel(const short* __restrict__ A,  // [16][32]
                                  const short* __restrict__ Bm, // [32][16]
                                  float* __restrict__ D) {      // [16][16]
  const int lane = threadIdx.x & (WAVE - 1);
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = *reinterpret_cast<const __bf16*>(A + (lane & 15) * 32 + (lane >> 4) * 8 + j);
    b[j] = *reinterpret_cast<const __bf16*>(Bm + ((lane >> 4) * 8 + j) * 16 + (lane & 15));
  }
  f32x4 c = {0, 0, 0, 0};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 4; ++reg)
    D[((lane >> 4) * 4 + reg) * 16 + (lane & 15)] = c[reg];
}

extern "C" void launch_mfma_probe(const void* A, const void* Bm, float* D,
                                  hipStream_t stream) {
  mfma_probe_kernel<<<dim3(1), dim3(64), 0, stream>>>(
      (const short*)A, (const short*)Bm, D);
}
