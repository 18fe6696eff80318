// Grouped (gather-style) expert MLP kernels — bf16 experts.
// Decode regime: N tokens × top_k experts = P (token, expert) pairs,
// each a GEMV against one expert's stacked weights
// (the `switch_mlp` layout, /root/reference/shard/server/model/deepseek_v2.py:101-112).
//
// Kernel 1: h[p, I] = silu(gate_e · x_t) * (up_e · x_t)   (fused)
// Kernel 2: out[t, H] += w_p * (down_e · h_p)             (atomic fp32 scatter)
//
// Quantized experts go through the w4a16 gather GEMV (w4a16.hip) from
// Python with the same pair arrays.

#include "hip_common.h"

#define MG_BLOCK 256
#define MG_WAVES (MG_BLOCK / WAVE)

// Fused gate/up GEMV + SwiGLU.  grid = (row_tiles, P).
__global__ __launch_bounds__(MG_BLOCK) void moe_gateup_kernel(
    const short* __restrict__ x,        // [N, H]
    const short* __restrict__ gate_w,   // [E, I, H]
    const short* __restrict__ up_w,     // [E, I, H]
    short* __restrict__ h,              // [P, I]
    const int* __restrict__ pair_token, const int* __restrict__ pair_expert,
    int H, int I) {
  const int p = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* x_lds = reinterpret_cast<float*>(smem_raw);  // [H]
  const short* xr = x + (long)pair_token[p] * H;
  for (int i = tid; i < H; i += MG_BLOCK) x_lds[i] = bfbits2f(xr[i]);
  __syncthreads();

  const long ebase = (long)pair_expert[p] * I * H;
  for (int o = blockIdx.x * MG_WAVES + wid; o < I; o += gridDim.x * MG_WAVES) {
    const short* grow = gate_w + ebase + (long)o * H;
    const short* urow = up_w + ebase + (long)o * H;
    float gdot = 0.0f, udot = 0.0f;
    for (int d = lane * 4; d < H; d += WAVE * 4) {
      short4v gv = *reinterpret_cast<const short4v*>(grow + d);
      short4v uv = *reinterpret_cast<const short4v*>(urow + d);
      const float* xp = x_lds + d;
      gdot += bfbits2f(gv.x) * xp[0] + bfbits2f(gv.y) * xp[1] +
              bfbits2f(gv.z) * xp[2] + bfbits2f(gv.w) * xp[3];
      udot += bfbits2f(uv.x) * xp[0] + bfbits2f(uv.y) * xp[1] +
              bfbits2f(uv.z) * xp[2] + bfbits2f(uv.w) * xp[3];
    }
    gdot = wave_sum(gdot);
    udot = wave_sum(udot);
    if (lane == 0) {
      float a = gdot / (1.0f + __expf(-gdot));  // silu
      h[(long)p * I + o] = (short)__bfloat16_as_ushort(f2bf(a * udot));
    }
  }
}

// Down-proj GEMV + weighted atomic scatter into fp32 out.
__global__ __launch_bounds__(MG_BLOCK) void moe_down_kernel(
    const short* __restrict__ h,        // [P, I]
    const short* __restrict__ down_w,   // [E, H, I]
    float* __restrict__ out,            // [N, H] fp32 (pre-zeroed)
    const int* __restrict__ pair_token, const int* __restrict__ pair_expert,
    const float* __restrict__ pair_weight, int I, int H) {
  const int p = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* h_lds = reinterpret_cast<float*>(smem_raw);  // [I]
  const short* hr = h + (long)p * I;
  for (int i = tid; i < I; i += MG_BLOCK) h_lds[i] = bfbits2f(hr[i]);
  __syncthreads();

  const float pw = pair_weight[p];
  const long ebase = (long)pair_expert[p] * H * I;
  float* orow = out + (long)pair_token[p] * H;
  for (int o = blockIdx.x * MG_WAVES + wid; o < H; o += gridDim.x * MG_WAVES) {
    const short* drow = down_w + ebase + (long)o * I;
    float dot = 0.0f;
    for (int d = lane * 4; d < I; d += WAVE * 4) {
      short4v dv = *reinterpret_cast<const short4v*>(drow + d);
      const float* hp = h_lds + d;
      dot += bfbits2f(dv.x) * hp[0] + bfbits2f(dv.y) * hp[1] +
             bfbits2f(dv.z) * hp[2] + bfbits2f(dv.w) * hp[3];
    }
    dot = wave_sum(dot);
    if (lane == 0) atomicAdd(orow + o, pw * dot);
  }
}

extern "C" void launch_moe_gateup(const void* x, const void* gate_w,
                                  const void* up_w, void* h,
                                  const int* pair_token,
                                  const int* pair_expert, int P, int H, int I,
                                  hipStream_t stream) {
  size_t smem = (size_t)H * sizeof(float);
  int gx = (I + MG_WAVES - 1) / MG_WAVES;
  if (gx > 1024) gx = 1024;
  moe_gateup_kernel<<<dim3(gx, P), dim3(MG_BLOCK), smem, stream>>>(
      (const short*)x, (const short*)gate_w, (const short*)up_w, (short*)h,
      pair_token, pair_expert, H, I);
}

extern "C" void launch_moe_down(const void* h, const void* down_w, float* out,
                                const int* pair_token, const int* pair_expert,
                                const float* pair_weight, int P, int I, int H,
                                hipStream_t stream) {
  size_t smem = (size_t)I * sizeof(float);
  int gx = (H + MG_WAVES - 1) / MG_WAVES;
  if (gx > 1024) gx = 1024;
  moe_down_kernel<<<dim3(gx, P), dim3(MG_BLOCK), smem, stream>>>(
      (const short*)h, (const short*)down_w, out, pair_token, pair_expert,
      pair_weight, I, H);
}
