// MLX-affine w4a16 / w8a16 quantized matmul kernels.
//
// Weight layout (checkpoint-compatible, /root/reference/shard/utils.py:54-65):
//   w_q   [O, H*bits/32] uint32, little-endian nibbles/bytes
//   scales[O, H/gs], biases[O, H/gs]  (bf16)
//   w[o,i] = scales[o, i/gs] * q + biases[o, i/gs]
//
// Decode regime (M small) is a GEMV: one wave per output row,
// x staged in LDS as fp32 together with per-word partial sums so the
// affine bias term costs one fma per 8 weights:
//   dot = sum_words  s_g * (sum_j q_j x_j)  +  b_g * (sum_j x_j).
//
// The same kernel gathers per-(token, expert) pairs for the MoE path
// (stacked expert weights [E, O, H*bits/32]) via optional index arrays.
// Large-M falls back to dequant (below) + hipBLASLt GEMM from Python.

#include "hip_common.h"

#define QG_BLOCK 256
#define QG_WAVES (QG_BLOCK / WAVE)

// x [M, H] bf16; y [M, O] fp32-accum -> bf16 out (or accumulate float)
// pair_token / pair_expert: if non-null, row m of the launch maps to
// token pair_token[m] and weight base pair_expert[m] * O * (H*bits/32).
template <int BITS>
__global__ __launch_bounds__(QG_BLOCK) void w4a16_gemv_kernel(
    const short* __restrict__ x, const unsigned int* __restrict__ wq,
    const short* __restrict__ scales, const short* __restrict__ biases,
    short* __restrict__ y, int M, int O, int H, int gs,
    const int* __restrict__ pair_token, const int* __restrict__ pair_expert) {
  constexpr int PER_WORD = 32 / BITS;   // 8 (4-bit) or 4 (8-bit)
  constexpr unsigned MASK = (1u << BITS) - 1u;
  const int m = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int words_per_row = H / PER_WORD;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* x_lds = reinterpret_cast<float*>(smem_raw);        // [H]
  float* xs_lds = x_lds + H;                                 // [H/PER_WORD] word sums

  const int tok = pair_token ? pair_token[m] : m;
  const short* xr = x + (long)tok * H;
  for (int i = tid; i < H; i += QG_BLOCK) x_lds[i] = bfbits2f(xr[i]);
  __syncthreads();
  for (int w = tid; w < words_per_row; w += QG_BLOCK) {
    float s = 0.0f;
#pragma unroll
    for (int j = 0; j < PER_WORD; ++j) s += x_lds[w * PER_WORD + j];
    xs_lds[w] = s;
  }
  __syncthreads();

  const long wbase = pair_expert ? (long)pair_expert[m] * O * words_per_row : 0;
  const long sbase = pair_expert ? (long)pair_expert[m] * O * (H / gs) : 0;
  const int groups_per_row = H / gs;
  const int words_per_group = gs / PER_WORD;

  for (int o = blockIdx.x * QG_WAVES + wid; o < O; o += gridDim.x * QG_WAVES) {
    const unsigned int* wrow = wq + wbase + (long)o * words_per_row;
    const short* srow = scales + sbase + (long)o * groups_per_row;
    const short* brow = biases + sbase + (long)o * groups_per_row;
    float dot = 0.0f;
    for (int w = lane; w < words_per_row; w += WAVE) {
      unsigned int bits = wrow[w];
      const int g = w / words_per_group;
      float inner = 0.0f;
      const float* xp = x_lds + w * PER_WORD;
#pragma unroll
      for (int j = 0; j < PER_WORD; ++j)
        inner += (float)((bits >> (BITS * j)) & MASK) * xp[j];
      dot += bfbits2f(srow[g]) * inner + bfbits2f(brow[g]) * xs_lds[w];
    }
    dot = wave_sum(dot);
    if (lane == 0) y[(long)m * O + o] = (short)__bfloat16_as_ushort(f2bf(dot));
  }
}

extern "C" void launch_w4a16_gemv(const void* x, const void* wq,
                                  const void* scales, const void* biases,
                                  void* y, int M, int O, int H, int gs,
                                  int bits, const int* pair_token,
                                  const int* pair_expert, hipStream_t stream) {
  const int per_word = 32 / bits;
  size_t smem = (size_t)(H + H / per_word) * sizeof(float);
  int gx = (O + QG_WAVES - 1) / QG_WAVES;
  if (gx > 2048) gx = 2048;
  dim3 grid(gx, M);
  if (bits == 4)
    w4a16_gemv_kernel<4><<<grid, dim3(QG_BLOCK), smem, stream>>>(
        (const short*)x, (const unsigned int*)wq, (const short*)scales,
        (const short*)biases, (short*)y, M, O, H, gs, pair_token, pair_expert);
  else
    w4a16_gemv_kernel<8><<<grid, dim3(QG_BLOCK), smem, stream>>>(
        (const short*)x, (const unsigned int*)wq, (const short*)scales,
        (const short*)biases, (short*)y, M, O, H, gs, pair_token, pair_expert);
}

// ---------------------------------------------------------------------------
// Dequantize to bf16 (for the large-M path: dequant + hipBLASLt GEMM).
// One block per output row; vectorized u32 loads.
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
    // PER_WORD shorts = 16B (4-bit) or 8B (8-bit): one vector store
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
