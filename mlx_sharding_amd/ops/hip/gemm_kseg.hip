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
