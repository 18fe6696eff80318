// MLX-affine w4a16 / w8a16 quantized matmul kernels.
//
// Weight layout (checkpoint-compatible, /root/reference/shard/utils.py:54-65):
//   w_q   [O, H*bits/32] uint32, little-endian nibbles/bytes
//   w[o,i] = scales[o, i/gs] * q + biases[o, i/gs]
//
// Decode-regime GEMM (M <= 64 tokens): weights are the traffic, so each
// weight word must be read ONCE for ALL tokens.  Two families:
//  - w4a16_gemm_small_kernel: scalar, x staged in LDS in [MT x CH]
//    32 KiB chunks, 2 output rows per wave, per-token fp32 accumulators
//    statically indexed (guide §5.4 rule 20).  The affine bias folds as
//      dot = s_g * (sum_j q_j x_j) + b_g * (sum_j x_j).
//  - w4a16_mfma_kernel: MFMA 16x16x32 with grid.z token tiles and
//    split-K over grid.y (below).
// Large-M (prefill) uses dequant (bottom) + hipBLASLt GEMM from Python;
// with the dequant-residency cache on (ops/__init__.py) decode also
// routes through hipBLASLt and these kernels serve memory-tight mode.

#include "hip_common.h"

#define QK_BLOCK 256
#define QK_WAVES (QK_BLOCK / WAVE)
#define QK_ROWS 2  // output rows per wave (ILP across rows)

template <int BITS, int MT>
__global__ __launch_bounds__(QK_BLOCK) void w4a16_gemm_small_kernel(
    const short* __restrict__ x, const unsigned int* __restrict__ wq,
    const short* __restrict__ scales, const short* __restrict__ biases,
    short* __restrict__ y, int M, int O, int H, int gs) {
  constexpr int PER_WORD = 32 / BITS;        // 8 (4-bit) / 4 (8-bit)
  constexpr unsigned MASK = (1u << BITS) - 1u;
  constexpr int CH = 32768 / (MT * 2);       // chunk elems: LDS = MT*CH*2B = 32KB
  const int m0 = blockIdx.y * MT;
  const int mt = min(MT, M - m0);
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int words_per_row = H / PER_WORD;
  const int words_per_group = gs / PER_WORD;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  short* x_lds = reinterpret_cast<short*>(smem_raw);  // [MT][CH]

  // --- stage the whole chunked x tile once (H <= CH case is one chunk) ---
  // Chunked loop restructured: rows are looped INSIDE the block so the
  // staging cost amortizes over many rows (each block owns a wide strip
  // of output rows); x chunks re-staged per chunk only.
  for (int c0 = 0; c0 < H; c0 += CH) {
    const int clen = min(CH, H - c0);
    __syncthreads();
    // vectorized staging: short4 loads, one token row at a time
    for (int t = 0; t < MT; ++t) {
      short4v* dst = reinterpret_cast<short4v*>(x_lds + t * CH);
      if (t < mt) {
        const short4v* src =
            reinterpret_cast<const short4v*>(x + (long)(m0 + t) * H + c0);
        for (int i = tid; i < clen / 4; i += QK_BLOCK) dst[i] = src[i];
        for (int i = clen / 4 + tid; i < CH / 4; i += QK_BLOCK)
          dst[i] = short4v{0, 0, 0, 0};
      } else {
        for (int i = tid; i < CH / 4; i += QK_BLOCK) dst[i] = short4v{0, 0, 0, 0};
      }
    }
    __syncthreads();

    const int w_lo = c0 / PER_WORD;
    const int w_hi = (c0 + clen) / PER_WORD;
    for (int row0 = (blockIdx.x * QK_WAVES + wid) * QK_ROWS; row0 < O;
         row0 += gridDim.x * QK_WAVES * QK_ROWS) {
      float dot[QK_ROWS][MT];
#pragma unroll
      for (int r = 0; r < QK_ROWS; ++r)
#pragma unroll
        for (int t = 0; t < MT; ++t) dot[r][t] = 0.0f;
#pragma unroll
      for (int r = 0; r < QK_ROWS; ++r) {
        const int o = row0 + r;
        if (o >= O) continue;
        const unsigned int* wrow = wq + (long)o * words_per_row;
        const short* srow = scales + (long)o * (H / gs);
        const short* brow = biases + (long)o * (H / gs);
        // 8 B/lane packed-word loads; both words share one quant group
        // (launch checks gs); word loop outer, tokens inner (VGPR trap).
        for (int w = w_lo + lane * 2; w < w_hi; w += WAVE * 2) {
          const uint2 wv = *reinterpret_cast<const uint2*>(wrow + w);
          const int g = w / words_per_group;
          const float sg = bfbits2f(srow[g]);
          const float bg = bfbits2f(brow[g]);
          const unsigned int wrds[2] = {wv.x, wv.y};
          const int dloc = w * PER_WORD - c0;
#pragma unroll
          for (int c = 0; c < 2; ++c) {
            const unsigned int bits = wrds[c];
#pragma unroll
            for (int t = 0; t < MT; ++t) {
              float inner = 0.0f, xsum = 0.0f;
              const short4v* xp = reinterpret_cast<const short4v*>(
                  x_lds + t * CH + dloc + c * PER_WORD);
#pragma unroll
              for (int v4 = 0; v4 < PER_WORD / 4; ++v4) {
                short4v xv = xp[v4];
                float x0 = bfbits2f(xv.x), x1 = bfbits2f(xv.y),
                      x2 = bfbits2f(xv.z), x3 = bfbits2f(xv.w);
                inner += (float)((bits >> (BITS * (v4 * 4 + 0))) & MASK) * x0 +
                         (float)((bits >> (BITS * (v4 * 4 + 1))) & MASK) * x1 +
                         (float)((bits >> (BITS * (v4 * 4 + 2))) & MASK) * x2 +
                         (float)((bits >> (BITS * (v4 * 4 + 3))) & MASK) * x3;
                xsum += x0 + x1 + x2 + x3;
              }
              dot[r][t] += sg * inner + bg * xsum;
            }
          }
        }
      }
      // partial write: accumulate chunks via atomic-free add (first
      // chunk writes, later chunks add) — avoided by keeping one chunk
      // for H <= CH; multi-chunk uses fp32 scratch accumulation below.
#pragma unroll
      for (int r = 0; r < QK_ROWS; ++r) {
        const int o = row0 + r;
        if (o >= O) continue;
#pragma unroll
        for (int t = 0; t < MT; ++t) {
          float v = wave_sum(dot[r][t]);
          if (lane == 0 && t < mt) {
            short* yp = y + (long)(m0 + t) * O + o;
            if (c0 == 0)
              *yp = (short)__bfloat16_as_ushort(f2bf(v));
            else
              *yp = (short)__bfloat16_as_ushort(f2bf(bfbits2f(*yp) + v));
          }
        }
      }
    }
  }
}

extern "C" void launch_w4a16_gemv(const void* x, const void* wq,
                                  const void* scales, const void* biases,
                                  void* y, int M, int O, int H, int gs,
                                  int bits, hipStream_t stream) {
  const int rows_per_block = QK_WAVES * QK_ROWS;
  int gx = (O + rows_per_block - 1) / rows_per_block;
  if (gx > 256) gx = 256;  // loop rows inside the block: amortize x staging
  const size_t smem = 32768;
#define QK_CASE(BB, TT)                                                      \
  w4a16_gemm_small_kernel<BB, TT>                                            \
      <<<dim3(gx, (M + TT - 1) / TT), dim3(QK_BLOCK), smem, stream>>>(       \
          (const short*)x, (const unsigned int*)wq, (const short*)scales,    \
          (const short*)biases, (short*)y, M, O, H, gs)
  if (bits == 4) QK_CASE(4, 8);
  else QK_CASE(8, 8);
#undef QK_CASE
}

// ---------------------------------------------------------------------------
// Dequantize to bf16 (for the large-M path: dequant + hipBLASLt GEMM).
// ---------------------------------------------------------------------------

template <int BITS>
__global__ void dequant_kernel(const unsigned int* __restrict__ wq,
                               const short* __restrict__ scales,
                               const short* __restrict__ biases,
                               short* __restrict__ out, int O, int H, int gs) {
  constexpr int PER_WORD = 32 / BITS;
  constexpr unsigned MASK = (1u << BITS) - 1u;
  const int o = blockIdx.x;
  const int words_per_row = H / PER_WORD;
  const int words_per_group = gs / PER_WORD;
  const unsigned int* wrow = wq + (long)o * words_per_row;
  const short* srow = scales + (long)o * (H / gs);
  const short* brow = biases + (long)o * (H / gs);
  short* orow = out + (long)o * H;
  for (int w = threadIdx.x; w < words_per_row; w += blockDim.x) {
    unsigned int bits = wrow[w];
    const int g = w / words_per_group;
    const float s = bfbits2f(srow[g]);
    const float b = bfbits2f(brow[g]);
    short vals[PER_WORD];
#pragma unroll
    for (int j = 0; j < PER_WORD; ++j)
      vals[j] = (short)__bfloat16_as_ushort(
          f2bf(s * (float)((bits >> (BITS * j)) & MASK) + b));
    if (PER_WORD == 8)
      *reinterpret_cast<short8v*>(orow + w * PER_WORD) =
          *reinterpret_cast<short8v*>(vals);
    else
      *reinterpret_cast<short4v*>(orow + w * PER_WORD) =
          *reinterpret_cast<short4v*>(vals);
  }
}

extern "C" void launch_dequant(const void* wq, const void* scales,
                               const void* biases, void* out, long O, int H,
                               int gs, int bits, hipStream_t stream) {
  dim3 grid((unsigned)O);
  if (bits == 4)
    dequant_kernel<4><<<grid, dim3(256), 0, stream>>>(
        (const unsigned int*)wq, (const short*)scales, (const short*)biases,
        (short*)out, (int)O, H, gs);
  else
    dequant_kernel<8><<<grid, dim3(256), 0, stream>>>(
        (const unsigned int*)wq, (const short*)scales, (const short*)biases,
        (short*)out, (int)O, H, gs);
}

// ---------------------------------------------------------------------------
// MFMA w4a16 GEMM for the decode regime (M <= 32 tokens per tile).
//
// y[M, O] = x[M, H] @ dequant(W)^T on mfma_f32_16x16x32_bf16:
//   A = 16 W-rows x 32 k  (dequantized nibbles -> bf16 fragments: lane
//       row = l&15, k = (l>>4)*8+j, i.e. ONE uint32 word per lane per
//       k-slice — the packed layout IS the fragment layout)
//   B = x^T: lane col = token = l&15, k = (l>>4)*8+j — a contiguous
//       16-byte read from the staged x row.
// Weights are read and dequantized ONCE for all 32 tokens (the scalar
// GEMV re-dequantized per token tile: 8x the VALU).  C accumulators
// persist across k-chunks so x is staged in 32 KiB LDS chunks with no
// partial-sum rounding.
// ---------------------------------------------------------------------------

typedef __bf16 w4bf16x8 __attribute__((ext_vector_type(8)));
typedef float w4f32x4 __attribute__((ext_vector_type(4)));

#define QM_WAVES 4
#define QM_BLOCK (QM_WAVES * WAVE)
#define QM_MTOK 32
#define QM_RT 1    // 16-row W tiles per wave (2 was ~neutral: fixed cost dominates)
#define QM_CH 512  // k-chunk elems
// +8 shorts of row padding: at stride 512*2B=1024B every lane of a 16-lane
// ds_read_b128 group lands on the SAME bank (16-way conflict, guide G4).
#define QM_LDS (QM_CH + 8)

// Split-K for small-O GEMVs: O/64 row-tiles alone leave the chip
// nearly idle (o_proj at O=2048 = 32 blocks on 256 CUs), so gridDim.y
// splits the k-chunks; split blocks accumulate into an fp32 scratch
// with atomics and a tiny convert kernel produces bf16 y.  gridDim.z
// tiles the tokens (one launch for any M).
template <int BITS>
__global__ __launch_bounds__(QM_BLOCK) void w4a16_mfma_kernel(
    const short* __restrict__ x, const unsigned int* __restrict__ wq,
    const short* __restrict__ scales, const short* __restrict__ biases,
    short* __restrict__ y, float* __restrict__ yf, int M, int O, int H,
    int gs) {
  constexpr int PER_WORD = 32 / BITS;
  constexpr unsigned MASK = (1u << BITS) - 1u;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int words_per_row = H / PER_WORD;
  const int m0 = blockIdx.z * QM_MTOK;
  const int mt = min(QM_MTOK, M - m0);
  // k-range of this split (chunk-aligned)
  const int ncz = (H + QM_CH - 1) / QM_CH;
  const int ncpb = (ncz + gridDim.y - 1) / gridDim.y;
  const int c_lo = blockIdx.y * ncpb * QM_CH;
  const int c_hi = min(H, c_lo + ncpb * QM_CH);

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  short* x_lds = reinterpret_cast<short*>(smem_raw);  // [QM_MTOK][QM_LDS]

  // QM_RT row-tiles per wave: the staged x tile (32 tok x CH x 2 B)
  // outweighs the nibble weights a 16-row tile reads 8:1 — wider row
  // strips amortize the staging traffic and LDS reads.
  const int row0 = (blockIdx.x * QM_WAVES + wid) * 16 * QM_RT;

  w4f32x4 acc0[QM_RT], acc1[QM_RT];
#pragma unroll
  for (int r = 0; r < QM_RT; ++r) {
    acc0[r] = w4f32x4{0, 0, 0, 0};
    acc1[r] = w4f32x4{0, 0, 0, 0};
  }
  const unsigned int* wrow[QM_RT];
  const short* srow[QM_RT];
  const short* brow[QM_RT];
#pragma unroll
  for (int r = 0; r < QM_RT; ++r) {
    const int wrow_r = min(row0 + r * 16 + (lane & 15), O - 1);
    wrow[r] = wq + (long)wrow_r * words_per_row;
    srow[r] = scales + (long)wrow_r * (H / gs);
    brow[r] = biases + (long)wrow_r * (H / gs);
  }

  for (int c0 = c_lo; c0 < c_hi; c0 += QM_CH) {
    const int clen = min(QM_CH, H - c0);
    __syncthreads();
#pragma unroll 4
    for (int t = 0; t < QM_MTOK; ++t) {
      short4v* dst = reinterpret_cast<short4v*>(x_lds + t * QM_LDS);
      if (t < mt) {
        const short4v* src =
            reinterpret_cast<const short4v*>(x + (long)(m0 + t) * H + c0);
        for (int i = threadIdx.x; i < clen / 4; i += QM_BLOCK) dst[i] = src[i];
        // zero the LDS tail: clamped-address prefetch slices past clen
        // multiply these columns (must not be stale)
        for (int i = clen / 4 + threadIdx.x; i < QM_CH / 4; i += QM_BLOCK)
          dst[i] = short4v{0, 0, 0, 0};
      } else {
        for (int i = threadIdx.x; i < QM_CH / 4; i += QM_BLOCK)
          dst[i] = short4v{0, 0, 0, 0};
      }
    }
    __syncthreads();
    if (row0 >= O) continue;  // keep barrier participation

    // prefetch this lane's weight words + scales for the whole chunk so
    // all global loads issue back-to-back (one latency per chunk, not
    // one per k-slice)
    constexpr int NSL = QM_CH / 32;           // k-slices per chunk
    unsigned int wbuf[QM_RT][NSL * (BITS == 4 ? 1 : 2)];
    short sraw[QM_RT][NSL], braw[QM_RT][NSL];  // raw bf16: converting at
                                               // load-site would force a
                                               // vmcnt wait per load
    // unconditional loads with a clamped address (guide §5 trap 4(c):
    // per-element load-or-zero selects serialize into vmcnt(0) chains);
    // tail slices load garbage that multiplies ZEROED x columns.
    const int kk_max = H - 8;
#pragma unroll
    for (int r = 0; r < QM_RT; ++r)
#pragma unroll
      for (int i = 0; i < NSL; ++i) {
        const int kk = min(c0 + i * 32 + (lane >> 4) * 8, kk_max);
        if (BITS == 4) {
          wbuf[r][i] = wrow[r][kk / 8];
        } else {
          wbuf[r][i * 2] = wrow[r][kk / 4];
          wbuf[r][i * 2 + 1] = wrow[r][kk / 4 + 1];
        }
        sraw[r][i] = srow[r][kk / gs];
        braw[r][i] = brow[r][kk / gs];
      }
    // fixed trip count: a runtime break rolls the loop and the compiler
    // sinks the prefetched loads back to their use sites (observed: 16
    // serial global loads per chunk); tail slices see zeroed x.
#pragma unroll
    for (int i = 0; i < NSL; ++i) {
      const int k0 = i * 32;
      // B fragments: x^T halves (tokens 0-15, 16-31) — shared by all
      // QM_RT row tiles
      const short* xb = x_lds + k0 + (lane >> 4) * 8;
      w4bf16x8 bf0 = *reinterpret_cast<const w4bf16x8*>(xb + (lane & 15) * QM_LDS);
      w4bf16x8 bf1 =
          *reinterpret_cast<const w4bf16x8*>(xb + ((lane & 15) + 16) * QM_LDS);
#pragma unroll
      for (int r = 0; r < QM_RT; ++r) {
        w4bf16x8 af;
        const float sg = bfbits2f(sraw[r][i]);
        const float bg = bfbits2f(braw[r][i]);
        if (BITS == 4) {
          const unsigned int bits = wbuf[r][i];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            af[j] = (__bf16)(sg * (float)((bits >> (4 * j)) & MASK) + bg);
        } else {
          const unsigned int b0 = wbuf[r][i * 2];
          const unsigned int b1 = wbuf[r][i * 2 + 1];
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            af[j] = (__bf16)(sg * (float)((b0 >> (8 * j)) & MASK) + bg);
            af[4 + j] = (__bf16)(sg * (float)((b1 >> (8 * j)) & MASK) + bg);
          }
        }
        acc0[r] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf0, acc0[r], 0, 0, 0);
        acc1[r] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf1, acc1[r], 0, 0, 0);
      }
    }
  }

  // epilogue: D[row=W-row, col=token]; lane writes 4 rows x 1 token per half
  if (row0 >= O) return;
#pragma unroll
  for (int r = 0; r < QM_RT; ++r)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int o = row0 + r * 16 + (lane >> 4) * 4 + reg;
      if (o < O) {
        const int t0 = lane & 15;
        if (yf != nullptr) {  // split-K: fp32 atomic partials
          if (t0 < mt) atomicAdd(yf + (long)(m0 + t0) * O + o, acc0[r][reg]);
          if (t0 + 16 < mt)
            atomicAdd(yf + (long)(m0 + t0 + 16) * O + o, acc1[r][reg]);
        } else {
          if (t0 < mt)
            y[(long)(m0 + t0) * O + o] =
                (short)__bfloat16_as_ushort(f2bf(acc0[r][reg]));
          if (t0 + 16 < mt)
            y[(long)(m0 + t0 + 16) * O + o] =
                (short)__bfloat16_as_ushort(f2bf(acc1[r][reg]));
        }
      }
    }
}

// ---------------------------------------------------------------------------
// Dense bf16 MFMA GEMV for the decode regime (M <= 64): y = x @ W^T.
//
// hipBLASLt's tall-skinny M=64 selections run ~1 TB/s on these shapes;
// this streams W at HBM rate.  Same fragment scheme as the MoE MFMA
// kernels (LDS-free: B-fragments straight from the L2-resident token
// rows), same split-K-over-grid.y + fp32-atomic machinery as the w4
// MFMA kernel for small-O shapes.  MZ = 16-token groups (template).
// ---------------------------------------------------------------------------

typedef __bf16 dgbf16x8 __attribute__((ext_vector_type(8)));

__global__ void f32_to_bf16_kernel(const float* __restrict__ src,
                                   short* __restrict__ dst, long n);

#define DG_WAVES 4
#define DG_BLOCK (DG_WAVES * WAVE)
#define DG_NSL 4

template <int MZ>
__global__ __launch_bounds__(DG_BLOCK) void bf16_gemv_mfma_kernel(
    const short* __restrict__ x,  // [M, H]
    const short* __restrict__ w,  // [O, H]
    short* __restrict__ y,        // [M, O]
    float* __restrict__ yf,       // [M, O] fp32 (split-K), or null
    int M, int O, int H) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int row0 = (blockIdx.x * DG_WAVES + wid) * 16;
  if (row0 >= O) return;
  const int wr = min(row0 + (lane & 15), O - 1);
  const short* wrow = w + (long)wr * H;
  const short* xrow[MZ];
#pragma unroll
  for (int z = 0; z < MZ; ++z)
    xrow[z] = x + (long)min(z * 16 + (lane & 15), M - 1) * H;

  w4f32x4 acc[MZ];
#pragma unroll
  for (int z = 0; z < MZ; ++z) acc[z] = w4f32x4{0, 0, 0, 0};

  // k-slices of this split (H % 32 == 0 enforced by the binding)
  const int nsl_total = H / 32;
  const int nspb = (nsl_total + gridDim.y - 1) / gridDim.y;
  const int sl_lo = blockIdx.y * nspb;
  const int sl_hi = min(nsl_total, sl_lo + nspb);

  int sl = sl_lo;
  for (; sl + DG_NSL <= sl_hi; sl += DG_NSL) {
    dgbf16x8 af[DG_NSL], bf[MZ][DG_NSL];
#pragma unroll
    for (int i = 0; i < DG_NSL; ++i) {
      const int koff = (sl + i) * 32 + (lane >> 4) * 8;
      af[i] = *reinterpret_cast<const dgbf16x8*>(wrow + koff);
#pragma unroll
      for (int z = 0; z < MZ; ++z)
        bf[z][i] = *reinterpret_cast<const dgbf16x8*>(xrow[z] + koff);
    }
#pragma unroll
    for (int i = 0; i < DG_NSL; ++i)
#pragma unroll
      for (int z = 0; z < MZ; ++z)
        acc[z] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bf[z][i],
                                                         acc[z], 0, 0, 0);
  }
  for (; sl < sl_hi; ++sl) {  // tail slices, once
    const int koff = sl * 32 + (lane >> 4) * 8;
    dgbf16x8 af = *reinterpret_cast<const dgbf16x8*>(wrow + koff);
#pragma unroll
    for (int z = 0; z < MZ; ++z) {
      dgbf16x8 bf = *reinterpret_cast<const dgbf16x8*>(xrow[z] + koff);
      acc[z] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[z], 0, 0, 0);
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

extern "C" void launch_bf16_gemv_mfma(const void* x, const void* w, void* y,
                                      float* yf, int nk, int M, int O, int H,
                                      hipStream_t stream) {
  const int gx = (O + DG_WAVES * 16 - 1) / (DG_WAVES * 16);
  const int mz = (M + 15) / 16;
  dim3 grid((unsigned)gx, (unsigned)nk);
  if (nk > 1) {
    (void)hipMemsetAsync(yf, 0, (size_t)M * O * sizeof(float), stream);
  }
  float* yfp = nk > 1 ? yf : nullptr;
#define DG_CASE(Z)                                                           \
  case Z:                                                                    \
    bf16_gemv_mfma_kernel<Z><<<grid, dim3(DG_BLOCK), 0, stream>>>(           \
        (const short*)x, (const short*)w, (short*)y, yfp, M, O, H);          \
    break;
  switch (mz) {
    DG_CASE(1)
    DG_CASE(2)
    DG_CASE(3)
    DG_CASE(4)
    default:
      break;
  }
#undef DG_CASE
  if (nk > 1) {
    const long n = (long)M * O;
    f32_to_bf16_kernel<<<dim3((unsigned)((n + 255) / 256)), dim3(256), 0,
                         stream>>>(yf, (short*)y, n);
  }
}

// ---------------------------------------------------------------------------
// LDS-tiled dense bf16 GEMM for the decode regime (M <= 64), deep-k
// shapes.  The LDS-free bf16_gemv above re-reads the x rows from L2 per
// (wave, k-slice) — ~4x the weight traffic at M=64 — and hipBLASLt caps
// at ~3.4 TB/s on [8192, 28672]-like shapes.  Here ALL 64 token rows
// stage through LDS once per 256-elem k-chunk (32 KB), each wave owns
// 16 W rows, weights stream ONCE at 16 B/lane, and B fragments are
// b128 LDS reads.  Split-K over grid.y with fp32 atomics (as the w4
// MFMA kernel).
// ---------------------------------------------------------------------------

#define DM_WAVES 4
#define DM_BLOCK (DM_WAVES * WAVE)
#define DM_CH 256
#define DM_LDS (DM_CH + 8)  // row pad: stagger LDS banks, keep 16B align

// R = W-row-tiles per wave: each x-chunk stage feeds R*16 weight rows
// per wave (staging amortization — at R=1 the x LDS traffic equalled
// the weight traffic and the kernel measured 0.9 TB/s)
template <int R>
__global__ __launch_bounds__(DM_BLOCK) void bf16_gemm_m64_kernel(
    const short* __restrict__ x,  // [M, H]
    const short* __restrict__ w,  // [O, H]
    short* __restrict__ y,        // [M, O]
    float* __restrict__ yf,       // fp32 split-K partials or null
    int M, int O, int H) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  int row0[R];
  const short* wrow[R];
#pragma unroll
  for (int r = 0; r < R; ++r) {
    row0[r] = ((blockIdx.x * DM_WAVES + wid) * R + r) * 16;
    wrow[r] = w + (long)min(row0[r] + (lane & 15), O - 1) * H;
  }

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  short* x_lds = reinterpret_cast<short*>(smem_raw);  // [64][DM_LDS]

  // k-range of this split (chunk-aligned)
  const int ncz = (H + DM_CH - 1) / DM_CH;
  const int ncpb = (ncz + gridDim.y - 1) / gridDim.y;
  const int c_lo = blockIdx.y * ncpb * DM_CH;
  const int c_hi = min(H, c_lo + ncpb * DM_CH);

  w4f32x4 acc[R][4];
#pragma unroll
  for (int r = 0; r < R; ++r)
#pragma unroll
    for (int z = 0; z < 4; ++z) acc[r][z] = w4f32x4{0, 0, 0, 0};

  const int kk_max = H - 8;
  for (int c0 = c_lo; c0 < c_hi; c0 += DM_CH) {
    const int clen = min(DM_CH, H - c0);
    __syncthreads();
    // flattened staging: all 256 threads, 16 short4v each (a per-row
    // loop used only 64 threads across 64 serial iterations)
    {
      constexpr int ROW4 = DM_CH / 4;
      for (int idx = threadIdx.x; idx < 64 * ROW4; idx += DM_BLOCK) {
        const int t = idx / ROW4;
        const int i = idx - t * ROW4;
        short4v v = short4v{0, 0, 0, 0};
        if (t < M && i * 4 < clen)
          v = *reinterpret_cast<const short4v*>(x + (long)t * H + c0 + i * 4);
        reinterpret_cast<short4v*>(x_lds + t * DM_LDS)[i] = v;
      }
    }
    __syncthreads();

    constexpr int NSL = DM_CH / 32;  // 8 k-slices per chunk
#pragma unroll
    for (int r = 0; r < R; ++r) {
      if (row0[r] >= O) continue;  // guard, not break: keep the unroll
      // batch this row-tile's A loads (clamped tail addresses read
      // garbage that multiplies zeroed x columns)
      dgbf16x8 af[NSL];
#pragma unroll
      for (int i = 0; i < NSL; ++i) {
        const int kk = min(c0 + i * 32 + (lane >> 4) * 8, kk_max);
        af[i] = *reinterpret_cast<const dgbf16x8*>(wrow[r] + kk);
      }
#pragma unroll
      for (int i = 0; i < NSL; ++i) {
        const short* xb = x_lds + i * 32 + (lane >> 4) * 8;
#pragma unroll
        for (int z = 0; z < 4; ++z) {
          dgbf16x8 bf = *reinterpret_cast<const dgbf16x8*>(
              xb + (z * 16 + (lane & 15)) * DM_LDS);
          acc[r][z] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bf,
                                                              acc[r][z],
                                                              0, 0, 0);
        }
      }
    }
  }

#pragma unroll
  for (int r = 0; r < R; ++r) {
    if (row0[r] >= O) continue;
#pragma unroll
    for (int z = 0; z < 4; ++z) {
      const int t = z * 16 + (lane & 15);
      if (t >= M) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int o = row0[r] + (lane >> 4) * 4 + reg;
        if (o < O) {
          if (yf != nullptr)
            atomicAdd(yf + (long)t * O + o, acc[r][z][reg]);
          else
            y[(long)t * O + o] =
                (short)__bfloat16_as_ushort(f2bf(acc[r][z][reg]));
        }
      }
    }
  }
}

#define DM_R 4  // row-tiles per wave (x staging amortized 1:4)

extern "C" int bf16_gemm_m64_nsplit(int M, int O, int H) {
  const int gx = (O + DM_WAVES * 16 * DM_R - 1) / (DM_WAVES * 16 * DM_R);
  const int ncz = (H + DM_CH - 1) / DM_CH;
  int nk = 2048 / (gx > 0 ? gx : 1);
  if (nk > ncz) nk = ncz;
  if (nk < 1) nk = 1;
  return nk;
}

extern "C" void launch_bf16_gemm_m64(const void* x, const void* w, void* y,
                                     float* yf, int nk, int M, int O, int H,
                                     hipStream_t stream) {
  const int gx = (O + DM_WAVES * 16 * DM_R - 1) / (DM_WAVES * 16 * DM_R);
  const size_t smem = (size_t)64 * DM_LDS * sizeof(short);
  if (nk > 1)
    (void)hipMemsetAsync(yf, 0, (size_t)M * O * sizeof(float), stream);
  float* yfp = nk > 1 ? yf : nullptr;
  bf16_gemm_m64_kernel<DM_R><<<dim3((unsigned)gx, (unsigned)nk),
                               dim3(DM_BLOCK), smem, stream>>>(
      (const short*)x, (const short*)w, (short*)y, yfp, M, O, H);
  if (nk > 1) {
    const long n = (long)M * O;
    f32_to_bf16_kernel<<<dim3((unsigned)((n + 255) / 256)), dim3(256), 0,
                         stream>>>(yf, (short*)y, n);
  }
}

extern "C" int bf16_gemv_nsplit(int M, int O, int H) {
  const int gx = (O + DG_WAVES * 16 - 1) / (DG_WAVES * 16);
  const int nslt = H / 32;
  int nk = 256 / (gx > 0 ? gx : 1);
  const int max_nk = (nslt + DG_NSL - 1) / DG_NSL;
  if (nk > max_nk) nk = max_nk;
  if (nk < 1) nk = 1;
  return nk;
}

__global__ void f32_to_bf16_kernel(const float* __restrict__ src,
                                   short* __restrict__ dst, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dst[i] = (short)__bfloat16_as_ushort(f2bf(src[i]));
}

extern "C" int w4a16_mfma_nsplit(int M, int O, int H) {
  // target >= 256 blocks; k-splits are QM_CH-chunk aligned
  const int gx = (O + QM_WAVES * 16 * QM_RT - 1) / (QM_WAVES * 16 * QM_RT);
  const int mz = (M + QM_MTOK - 1) / QM_MTOK;
  const int ncz = (H + QM_CH - 1) / QM_CH;
  int nk = 256 / (gx * mz > 0 ? gx * mz : 1);
  if (nk > ncz) nk = ncz;
  if (nk < 1) nk = 1;
  return nk;
}

extern "C" void launch_w4a16_mfma(const void* x, const void* wq,
                                  const void* scales, const void* biases,
                                  void* y, float* yf, int nk, int M, int O,
                                  int H, int gs, int bits,
                                  hipStream_t stream) {
  const int gx = (O + QM_WAVES * 16 * QM_RT - 1) / (QM_WAVES * 16 * QM_RT);
  const int mz = (M + QM_MTOK - 1) / QM_MTOK;
  const size_t smem = QM_MTOK * QM_LDS * sizeof(short);
  dim3 grid((unsigned)gx, (unsigned)nk, (unsigned)mz);
  if (nk > 1)
    (void)hipMemsetAsync(yf, 0, (size_t)M * O * sizeof(float), stream);
  float* yfp = nk > 1 ? yf : nullptr;
  if (bits == 4)
    w4a16_mfma_kernel<4><<<grid, dim3(QM_BLOCK), smem, stream>>>(
        (const short*)x, (const unsigned int*)wq, (const short*)scales,
        (const short*)biases, (short*)y, yfp, M, O, H, gs);
  else
    w4a16_mfma_kernel<8><<<grid, dim3(QM_BLOCK), smem, stream>>>(
        (const short*)x, (const unsigned int*)wq, (const short*)scales,
        (const short*)biases, (short*)y, yfp, M, O, H, gs);
  if (nk > 1) {
    const long n = (long)M * O;
    f32_to_bf16_kernel<<<dim3((unsigned)((n + 255) / 256)), dim3(256), 0,
                         stream>>>(yf, (short*)y, n);
  }
}
