// Fused elementwise / normalization kernels for the decode hot path.
// All memory-bound: bf16 I/O vectorized as short4/short8 (guide G13 —
// scalar bf16 loads cost ~2x), fp32 accumulation, grid-stride where the
// shape demands it.
#include "hip_common.h"

// ---------------------------------------------------------------------------
// RMSNorm: y = x / rms(x) * (w + w_off).  One block per row.
// Optional fused residual: h = x + r; y = norm(h); h written back.
// ---------------------------------------------------------------------------

template <int BLOCK, bool RESIDUAL>
__global__ void rms_norm_kernel(const short* __restrict__ x,
                                const short* __restrict__ resid,
                                short* __restrict__ y,
                                short* __restrict__ h_out,
                                const short* __restrict__ w, int H, float eps,
                                float w_off, long xstride) {
  __shared__ float scratch[BLOCK / WAVE];
  const long row = blockIdx.x;
  const short* xr = x + row * xstride;  // strided: x may be a fused-split view
  const short* rr = RESIDUAL ? resid + row * H : nullptr;
  short* yr = y + row * H;
  short* hr = RESIDUAL ? h_out + row * H : nullptr;

  // Accumulate sum of squares with vectorized loads.
  float ss = 0.0f;
  const int nvec = H / 4;
  const short4v* xv = reinterpret_cast<const short4v*>(xr);
  const short4v* rv = reinterpret_cast<const short4v*>(rr);
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    short4v v = xv[i];
    float f0 = bfbits2f(v.x), f1 = bfbits2f(v.y), f2 = bfbits2f(v.z),
          f3 = bfbits2f(v.w);
    if (RESIDUAL) {
      short4v u = rv[i];
      f0 += bfbits2f(u.x); f1 += bfbits2f(u.y);
      f2 += bfbits2f(u.z); f3 += bfbits2f(u.w);
      short4v hv;
      hv.x = (short)__bfloat16_as_ushort(f2bf(f0));
      hv.y = (short)__bfloat16_as_ushort(f2bf(f1));
      hv.z = (short)__bfloat16_as_ushort(f2bf(f2));
      hv.w = (short)__bfloat16_as_ushort(f2bf(f3));
      reinterpret_cast<short4v*>(hr)[i] = hv;
      // re-read rounded values so y is computed from the stored h
      f0 = bfbits2f(hv.x); f1 = bfbits2f(hv.y);
      f2 = bfbits2f(hv.z); f3 = bfbits2f(hv.w);
    }
    ss += f0 * f0 + f1 * f1 + f2 * f2 + f3 * f3;
  }
  ss = block_sum<BLOCK>(ss, scratch);
  const float inv = rsqrtf(ss / (float)H + eps);

  const short4v* wv = reinterpret_cast<const short4v*>(w);
  const short4v* src = RESIDUAL ? reinterpret_cast<const short4v*>(hr)
                                : xv;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    short4v v = src[i];
    short4v wvv = wv[i];
    short4v o;
    o.x = (short)__bfloat16_as_ushort(f2bf(bfbits2f(v.x) * inv * (bfbits2f(wvv.x) + w_off)));
    o.y = (short)__bfloat16_as_ushort(f2bf(bfbits2f(v.y) * inv * (bfbits2f(wvv.y) + w_off)));
    o.z = (short)__bfloat16_as_ushort(f2bf(bfbits2f(v.z) * inv * (bfbits2f(wvv.z) + w_off)));
    o.w = (short)__bfloat16_as_ushort(f2bf(bfbits2f(v.w) * inv * (bfbits2f(wvv.w) + w_off)));
    reinterpret_cast<short4v*>(yr)[i] = o;
  }
}

extern "C" void launch_rms_norm(const void* x, void* y, const void* w,
                                long rows, int H, float eps, float w_off,
                                long xstride, hipStream_t stream) {
  constexpr int BLOCK = 256;
  rms_norm_kernel<BLOCK, false><<<dim3((unsigned)rows), dim3(BLOCK), 0, stream>>>(
      (const short*)x, nullptr, (short*)y, nullptr, (const short*)w, H, eps,
      w_off, xstride);
}

extern "C" void launch_rms_norm_residual(const void* x, const void* resid,
                                         void* y, void* h_out, const void* w,
                                         long rows, int H, float eps,
                                         float w_off, hipStream_t stream) {
  constexpr int BLOCK = 256;
  rms_norm_kernel<BLOCK, true><<<dim3((unsigned)rows), dim3(BLOCK), 0, stream>>>(
      (const short*)x, (const short*)resid, (short*)y, (short*)h_out,
      (const short*)w, H, eps, w_off, H);
}

// ---------------------------------------------------------------------------
// SwiGLU / GeGLU: y = act(gate) * up, N elements. Grid-stride, short4.
// ---------------------------------------------------------------------------

template <bool GELU>
__global__ void glu_kernel(const short* __restrict__ gate,
                           const short* __restrict__ up,
                           short* __restrict__ y, long n4) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    short4v g = reinterpret_cast<const short4v*>(gate)[i];
    short4v u = reinterpret_cast<const short4v*>(up)[i];
    float gf[4] = {bfbits2f(g.x), bfbits2f(g.y), bfbits2f(g.z), bfbits2f(g.w)};
    float uf[4] = {bfbits2f(u.x), bfbits2f(u.y), bfbits2f(u.z), bfbits2f(u.w)};
    short4v o;
    short* op = reinterpret_cast<short*>(&o);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float a;
      if (GELU) {  // tanh approximation (gemma2)
        float x = gf[k];
        float inner = 0.7978845608028654f * (x + 0.044715f * x * x * x);
        a = 0.5f * x * (1.0f + tanhf(inner));
      } else {  // silu
        a = gf[k] / (1.0f + __expf(-gf[k]));
      }
      op[k] = (short)__bfloat16_as_ushort(f2bf(a * uf[k]));
    }
    reinterpret_cast<short4v*>(y)[i] = o;
  }
}

template <bool GELU>
__global__ void glu_strided_kernel(const short* __restrict__ gate,
                                   const short* __restrict__ up,
                                   short* __restrict__ y, long rows, int I4,
                                   long gstride, long ustride) {
  const long total = rows * I4;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const long r = i / I4;
    const int c = (int)(i - r * I4);
    short4v g = *reinterpret_cast<const short4v*>(gate + r * gstride + c * 4);
    short4v u = *reinterpret_cast<const short4v*>(up + r * ustride + c * 4);
    float gf[4] = {bfbits2f(g.x), bfbits2f(g.y), bfbits2f(g.z), bfbits2f(g.w)};
    float uf[4] = {bfbits2f(u.x), bfbits2f(u.y), bfbits2f(u.z), bfbits2f(u.w)};
    short4v o;
    short* op = reinterpret_cast<short*>(&o);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float a;
      if (GELU) {
        float x = gf[k];
        float inner = 0.7978845608028654f * (x + 0.044715f * x * x * x);
        a = 0.5f * x * (1.0f + tanhf(inner));
      } else {
        a = gf[k] / (1.0f + __expf(-gf[k]));
      }
      op[k] = (short)__bfloat16_as_ushort(f2bf(a * uf[k]));
    }
    reinterpret_cast<short4v*>(y)[i] = o;
  }
}

extern "C" void launch_glu_strided(const void* gate, const void* up, void* y,
                                   long rows, int I, long gstride,
                                   long ustride, bool gelu,
                                   hipStream_t stream) {
  const long n4 = rows * (I / 4);
  int block = 256;
  long want = (n4 + block - 1) / block;
  int grid = (int)min(want, (long)(256 * 8));
  if (grid < 1) grid = 1;
  if (gelu)
    glu_strided_kernel<true><<<dim3(grid), dim3(block), 0, stream>>>(
        (const short*)gate, (const short*)up, (short*)y, rows, I / 4, gstride,
        ustride);
  else
    glu_strided_kernel<false><<<dim3(grid), dim3(block), 0, stream>>>(
        (const short*)gate, (const short*)up, (short*)y, rows, I / 4, gstride,
        ustride);
}

extern "C" void launch_glu(const void* gate, const void* up, void* y, long n,
                           bool gelu, hipStream_t stream) {
  long n4 = n / 4;
  int block = 256;
  long want = (n4 + block - 1) / block;
  int grid = (int)min(want, (long)(256 * 8));
  if (grid < 1) grid = 1;
  if (gelu)
    glu_kernel<true><<<dim3(grid), dim3(block), 0, stream>>>(
        (const short*)gate, (const short*)up, (short*)y, n4);
  else
    glu_kernel<false><<<dim3(grid), dim3(block), 0, stream>>>(
        (const short*)gate, (const short*)up, (short*)y, n4);
}

// ---------------------------------------------------------------------------
// Softcap: y = cap * tanh(x / cap)
// ---------------------------------------------------------------------------

__global__ void softcap_kernel(const short* __restrict__ x, short* __restrict__ y,
                               long n4, float cap) {
  const long stride = (long)gridDim.x * blockDim.x;
  const float inv = 1.0f / cap;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    short4v v = reinterpret_cast<const short4v*>(x)[i];
    short4v o;
    o.x = (short)__bfloat16_as_ushort(f2bf(cap * tanhf(bfbits2f(v.x) * inv)));
    o.y = (short)__bfloat16_as_ushort(f2bf(cap * tanhf(bfbits2f(v.y) * inv)));
    o.z = (short)__bfloat16_as_ushort(f2bf(cap * tanhf(bfbits2f(v.z) * inv)));
    o.w = (short)__bfloat16_as_ushort(f2bf(cap * tanhf(bfbits2f(v.w) * inv)));
    reinterpret_cast<short4v*>(y)[i] = o;
  }
}

extern "C" void launch_softcap(const void* x, void* y, long n, float cap,
                               hipStream_t stream) {
  long n4 = n / 4;
  int block = 256;
  long want = (n4 + block - 1) / block;
  int grid = (int)min(want, (long)(256 * 8));
  if (grid < 1) grid = 1;
  softcap_kernel<<<dim3(grid), dim3(block), 0, stream>>>(
      (const short*)x, (short*)y, n4, cap);
}

// ---------------------------------------------------------------------------
// RoPE: x [rows=B*T, n_heads, D], cos/sin [T, D/2] fp32 (host-precomputed
// tables — guide Appendix B: never sinf/cosf per element on device).
// Half-split (llama) or interleaved (deepseek rope slice).
// One thread per (row, head, pair).
// ---------------------------------------------------------------------------

template <bool INTERLEAVED>
__global__ void rope_kernel(const short* __restrict__ x, short* __restrict__ y,
                            const float* __restrict__ cost,
                            const float* __restrict__ sint,
                            int T, int n_heads, int D, long xrstride,
                            int xhstride) {
  const int half = D / 2;
  const long bh = blockIdx.x;
  const long row = bh / n_heads;
  const int head = bh % n_heads;
  const int t = row % T;  // row = b*T + t
  // x may be a strided view (fused-qkv split / nope-rope slice)
  const short* xr = x + row * xrstride + (long)head * xhstride;
  short* yr = y + (row * n_heads + head) * D;
  const float* c = cost + (long)t * half;
  const float* s = sint + (long)t * half;
  for (int p = threadIdx.x; p < half; p += blockDim.x) {
    float x1, x2;
    int i1, i2;
    if (INTERLEAVED) {
      i1 = 2 * p;
      i2 = 2 * p + 1;
    } else {
      i1 = p;
      i2 = p + half;
    }
    x1 = bfbits2f(xr[i1]);
    x2 = bfbits2f(xr[i2]);
    float o1 = x1 * c[p] - x2 * s[p];
    float o2 = x2 * c[p] + x1 * s[p];
    yr[i1] = (short)__bfloat16_as_ushort(f2bf(o1));
    yr[i2] = (short)__bfloat16_as_ushort(f2bf(o2));
  }
}

extern "C" void launch_rope(const void* x, void* y, const float* cost,
                            const float* sint, long rows, int T, int n_heads,
                            int D, bool interleaved, long xrstride,
                            long xhstride, hipStream_t stream) {
  dim3 grid((unsigned)(rows * n_heads));
  int block = D / 2 < 64 ? 64 : (D / 2 > 256 ? 256 : D / 2);
  if (interleaved)
    rope_kernel<true><<<grid, dim3(block), 0, stream>>>(
        (const short*)x, (short*)y, cost, sint, T, n_heads, D, xrstride,
        (int)xhstride);
  else
    rope_kernel<false><<<grid, dim3(block), 0, stream>>>(
        (const short*)x, (short*)y, cost, sint, T, n_heads, D, xrstride,
        (int)xhstride);
}

// ---------------------------------------------------------------------------
// Fused MLA KV-cache append: scatter kv_b output + broadcast roped k_pe
// straight into the caches (replaces cat + head-expand + two index_copy
// launches per layer on the decode path).
// kvh  [B, T, nh, nope+vd]  (contiguous kv_b_proj output)
// kpe  [B, T, rope]         (roped shared key slice)
// kcache [B, nh, Scap, nope+rope]; vcache [B, nh, Scap, vd]
// position = *pos_ptr (graph mode) or pos0, plus the row's t.
// ---------------------------------------------------------------------------

__global__ void mla_append_kv_kernel(
    const short* __restrict__ kvh, const short* __restrict__ kpe,
    short* __restrict__ kcache, short* __restrict__ vcache,
    const int* __restrict__ pos_ptr, int pos0, int B, int T, int nh,
    int nope, int vd, int rope, long Scap, long kpe_rstride) {
  const int bh = blockIdx.x;
  const int t = blockIdx.y;
  const int b = bh / nh;
  const int h = bh % nh;
  const long pos = (pos_ptr ? *pos_ptr : pos0) + t;
  const short* src = kvh + (((long)b * T + t) * nh + h) * (nope + vd);
  const short* pe = kpe + ((long)b * T + t) * kpe_rstride;  // may be a slice view
  short* krow = kcache + (((long)b * nh + h) * Scap + pos) * (nope + rope);
  short* vrow = vcache + (((long)b * nh + h) * Scap + pos) * vd;
  for (int i = threadIdx.x; i < nope; i += blockDim.x) krow[i] = src[i];
  for (int i = threadIdx.x; i < rope; i += blockDim.x) krow[nope + i] = pe[i];
  for (int i = threadIdx.x; i < vd; i += blockDim.x) vrow[i] = src[nope + i];
}

// ---------------------------------------------------------------------------
// Fused GQA rope-k + KV-cache append: applies RoPE to the new k rows and
// scatters k/v straight into the caches — replaces a rope launch plus
// two index_copy launches per layer on the llama/gemma decode path.
// k/v [B, T, Hkv, D] (contiguous), caches [B, Hkv, Scap, D].
// ---------------------------------------------------------------------------

template <bool INTERLEAVED>
__global__ void rope_append_kv_kernel(
    const short* __restrict__ k, const short* __restrict__ v,
    const float* __restrict__ cost, const float* __restrict__ sint,
    short* __restrict__ kcache, short* __restrict__ vcache,
    const int* __restrict__ pos_ptr, int pos0, int B, int T, int Hkv, int D,
    long Scap, long k_rstride, long v_rstride) {
  const int bh = blockIdx.x;
  const int t = blockIdx.y;
  const int b = bh / Hkv;
  const int h = bh % Hkv;
  const long pos = (pos_ptr ? *pos_ptr : pos0) + t;
  // k/v may be fused-qkv split views: per-(b,t) row stride differs from
  // Hkv*D but heads stay contiguous within the slice
  const short* kr = k + ((long)b * T + t) * k_rstride + (long)h * D;
  const short* vr = v + ((long)b * T + t) * v_rstride + (long)h * D;
  short* kd = kcache + (((long)b * Hkv + h) * Scap + pos) * D;
  short* vd = vcache + (((long)b * Hkv + h) * Scap + pos) * D;
  const int half = D / 2;
  const float* c = cost + (long)t * half;
  const float* s = sint + (long)t * half;
  for (int p = threadIdx.x; p < half; p += blockDim.x) {
    int i1, i2;
    if (INTERLEAVED) {
      i1 = 2 * p;
      i2 = 2 * p + 1;
    } else {
      i1 = p;
      i2 = p + half;
    }
    const float x1 = bfbits2f(kr[i1]), x2 = bfbits2f(kr[i2]);
    kd[i1] = (short)__bfloat16_as_ushort(f2bf(x1 * c[p] - x2 * s[p]));
    kd[i2] = (short)__bfloat16_as_ushort(f2bf(x2 * c[p] + x1 * s[p]));
    vd[i1] = vr[i1];
    vd[i2] = vr[i2];
  }
}

extern "C" void launch_rope_append_kv(const void* k, const void* v,
                                      const float* cost, const float* sint,
                                      void* kcache, void* vcache,
                                      const int* pos_ptr, int pos0, int B,
                                      int T, int Hkv, int D, long Scap,
                                      bool interleaved, long k_rstride,
                                      long v_rstride, hipStream_t stream) {
  dim3 grid((unsigned)(B * Hkv), (unsigned)T);
  int block = D / 2 < 64 ? 64 : (D / 2 > 256 ? 256 : D / 2);
  if (interleaved)
    rope_append_kv_kernel<true><<<grid, dim3(block), 0, stream>>>(
        (const short*)k, (const short*)v, cost, sint, (short*)kcache,
        (short*)vcache, pos_ptr, pos0, B, T, Hkv, D, Scap, k_rstride,
        v_rstride);
  else
    rope_append_kv_kernel<false><<<grid, dim3(block), 0, stream>>>(
        (const short*)k, (const short*)v, cost, sint, (short*)kcache,
        (short*)vcache, pos_ptr, pos0, B, T, Hkv, D, Scap, k_rstride,
        v_rstride);
}

extern "C" void launch_mla_append_kv(const void* kvh, const void* kpe,
                                     void* kcache, void* vcache,
                                     const int* pos_ptr, int pos0, int B,
                                     int T, int nh, int nope, int vd, int rope,
                                     long Scap, long kpe_rstride,
                                     hipStream_t stream) {
  mla_append_kv_kernel<<<dim3((unsigned)(B * nh), (unsigned)T), dim3(128), 0,
                         stream>>>((const short*)kvh, (const short*)kpe,
                                   (short*)kcache, (short*)vcache, pos_ptr,
                                   pos0, B, T, nh, nope, vd, rope, Scap,
                                   kpe_rstride);
}
