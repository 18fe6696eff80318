// Python bindings for the gfx950 kernel library (torch extension).
// Pure HIP tree — no CUDA shims; built by hipcc via torch cpp_extension
// with PYTORCH_ROCM_ARCH=gfx950.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "hip_common.h"

extern "C" {
void launch_rms_norm(const void*, void*, const void*, long, int, float, float,
                     long, hipStream_t);
void launch_rms_norm_residual(const void*, const void*, void*, void*,
                              const void*, long, int, float, float,
                              hipStream_t);
void launch_glu(const void*, const void*, void*, long, bool, hipStream_t);
void launch_glu_strided(const void*, const void*, void*, long, int, long,
                        long, bool, hipStream_t);
void launch_softcap(const void*, void*, long, float, hipStream_t);
void launch_rope(const void*, void*, const float*, const float*, long, int,
                 int, int, bool, long, long, hipStream_t);
void launch_mla_append_kv(const void*, const void*, void*, void*, const int*,
                          int, int, int, int, int, int, int, long, long,
                          hipStream_t);
void launch_rope_append_kv(const void*, const void*, const float*,
                           const float*, void*, void*, const int*, int, int,
                           int, int, int, long, bool, long, long,
                           hipStream_t);
void launch_attn_decode(const void*, const void*, const void*, void*, float*,
                        float*, int, const int*, int, int, int, int, long,
                        int, int, long, float, float, int, hipStream_t);
bool attn_decode_supported_shape(int, int);
void launch_attn_prefill(const void*, const void*, const void*, void*, int,
                         int, int, int, int, long, long, int, int, float,
                         float, int, int, hipStream_t);
bool attn_prefill_supported(int, int);
void launch_mfma_probe(const void*, const void*, float*, hipStream_t);
void launch_w4a16_gemv(const void*, const void*, const void*, const void*,
                       void*, int, int, int, int, int, hipStream_t);
int w4a16_mfma_nsplit(int, int, int);
void launch_bf16_gemv_mfma(const void*, const void*, void*, float*, int, int,
                           int, int, hipStream_t);
int bf16_gemv_nsplit(int, int, int);
void launch_bf16_gemm_m64(const void*, const void*, void*, float*, int, int,
                          int, int, hipStream_t);
int bf16_gemm_m64_nsplit(int, int, int);
void launch_w4a16_mfma(const void*, const void*, const void*, const void*,
                       void*, float*, int, int, int, int, int, int,
                       hipStream_t);
void launch_dequant(const void*, const void*, const void*, void*, long, int,
                    int, int, hipStream_t);
void launch_moe_scatter_rows(const void*, void*, const int*, const int*, int,
                             int, hipStream_t);
void launch_moe_gather_reduce(const void*, const int*, const float*, void*,
                              int, int, int, hipStream_t);
void launch_moe_gateup_mfma(const void*, const void*, const void*, void*,
                            const int*, const int*, const int*, const int*,
                            int, int, int, hipStream_t);
void launch_moe_down_mfma(const void*, const void*, float*, const int*,
                          const int*, const int*, const int*, const float*,
                          int, int, int, hipStream_t);
void launch_moe_gateup_grouped(const void*, const void*, const void*, void*,
                               const int*, const int*, const int*, const int*,
                               int, int, int, hipStream_t);
void launch_moe_down_grouped(const void*, const void*, float*, const int*,
                             const int*, const int*, const int*, const float*,
                             int, int, int, hipStream_t);
void launch_moe_w4_mfma(const void*, const void*, const void*, const void*,
                        void*, const int*, const int*, const int*,
                        const int*, int, int, int, int, int, hipStream_t);
void launch_moe_gate_subranges(const void*, int*, float*, int*, int*, int*,
                               int, int, int, int, int, float, int,
                               hipStream_t);
void launch_moe_w4f16_gateup(const void*, const void*, const void*,
                             const void*, const void*, const void*,
                             const void*, void*, const int*, const int*,
                             const int*, const int*, int, int, int, int, int,
                             hipStream_t);
void launch_moe_w4f16_down(const void*, const void*, const void*, const void*,
                           float*, const int*, const int*, const int*,
                           const int*, const float*, int, int, int, int, int,
                           hipStream_t);
void launch_w4f16_gemv(const void*, const void*, const void*, const void*,
                       void*, float*, int, int, int, int, int, int,
                       hipStream_t);
int w4f16_gemv_nsplit(int, int, int);
void launch_gemm_kseg(const void*, const void*, void*, void*, int, int, int,
                      int, hipStream_t);
void launch_gemm_kseg_w4(const void*, const void*, const void*, const void*,
                         void*, void*, int, int, int, int, int, hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
}

// View t as [rows, last-dim] with a uniform row stride (a fused-output
// column-slice view qualifies) — lets kernels read split views in place
// instead of the binding materializing a .contiguous() copy per call.
bool row_view2(const torch::Tensor& t, long& rows, long& rstride) {
  const int d = t.dim();
  if (d < 2 || t.stride(d - 1) != 1) return false;
  if (d == 2) {
    rows = t.size(0);
    rstride = t.stride(0);
    return true;
  }
  if (d == 3) {
    if (t.stride(0) != t.size(1) * t.stride(1)) return false;
    rows = t.size(0) * t.size(1);
    rstride = t.stride(1);
    return true;
  }
  return false;
}

torch::Tensor rms_norm(torch::Tensor x, torch::Tensor w, double eps,
                       double w_off) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  const int H = x.size(-1);
  TORCH_CHECK(H % 4 == 0, "H must be divisible by 4");
  long rows, rstride;
  torch::Tensor xv = x;
  if (!row_view2(x, rows, rstride) || rstride % 4 != 0
      || ((uintptr_t)x.data_ptr()) % 8 != 0) {  // short4v-aligned rows
    xv = x.contiguous();
    rows = xv.numel() / H;
    rstride = H;
  }
  auto y = torch::empty(x.sizes(), x.options());
  launch_rms_norm(xv.data_ptr(), y.data_ptr(), w.contiguous().data_ptr(), rows,
                  H, (float)eps, (float)w_off, rstride, cur_stream());
  return y;
}

std::vector<torch::Tensor> rms_norm_residual(torch::Tensor x,
                                             torch::Tensor resid,
                                             torch::Tensor w, double eps,
                                             double w_off) {
  check_bf16(x, "x");
  auto xc = x.contiguous();
  auto rc = resid.contiguous();
  const int H = xc.size(-1);
  long rows = xc.numel() / H;
  auto y = torch::empty_like(xc);
  auto h = torch::empty_like(xc);
  launch_rms_norm_residual(xc.data_ptr(), rc.data_ptr(), y.data_ptr(),
                           h.data_ptr(), w.contiguous().data_ptr(), rows, H,
                           (float)eps, (float)w_off, cur_stream());
  return {y, h};
}

torch::Tensor glu(torch::Tensor gate, torch::Tensor up, bool gelu) {
  check_bf16(gate, "gate");
  TORCH_CHECK(gate.numel() == up.numel(), "gate/up size mismatch");
  TORCH_CHECK(gate.numel() % 4 == 0, "numel must be divisible by 4");
  const int I = gate.size(-1);
  long growz, gstride, urows, ustride;
  // fused gate|up split views read in place (no .contiguous() copies)
  if (!(gate.is_contiguous() && up.is_contiguous())
      && row_view2(gate, growz, gstride) && row_view2(up, urows, ustride)
      && growz == urows && up.size(-1) == I && I % 4 == 0
      && gstride % 4 == 0 && ustride % 4 == 0
      && ((uintptr_t)gate.data_ptr()) % 8 == 0
      && ((uintptr_t)up.data_ptr()) % 8 == 0) {
    auto y = torch::empty(gate.sizes(), gate.options());
    launch_glu_strided(gate.data_ptr(), up.data_ptr(), y.data_ptr(), growz, I,
                       gstride, ustride, gelu, cur_stream());
    return y;
  }
  auto g = gate.contiguous();
  auto u = up.contiguous();
  auto y = torch::empty_like(g);
  launch_glu(g.data_ptr(), u.data_ptr(), y.data_ptr(), g.numel(), gelu,
             cur_stream());
  return y;
}

torch::Tensor softcap_op(torch::Tensor x, double cap) {
  check_bf16(x, "x");
  auto xc = x.contiguous();
  TORCH_CHECK(xc.numel() % 4 == 0, "numel must be divisible by 4");
  auto y = torch::empty_like(xc);
  launch_softcap(xc.data_ptr(), y.data_ptr(), xc.numel(), (float)cap,
                 cur_stream());
  return y;
}

// x: [B, T, nH, D]; cos/sin: [T, D/2] fp32
torch::Tensor apply_rope(torch::Tensor x, torch::Tensor cos, torch::Tensor sin,
                         bool interleaved) {
  check_bf16(x, "x");
  TORCH_CHECK(x.dim() == 4, "x must be [B, T, nH, D]");
  auto cc = cos.contiguous();
  auto sc = sin.contiguous();
  TORCH_CHECK(cc.scalar_type() == torch::kFloat32, "cos must be fp32");
  const int B = x.size(0), T = x.size(1), nH = x.size(2), D = x.size(3);
  TORCH_CHECK(cc.size(0) == T && cc.size(1) == D / 2, "cos shape mismatch");
  // fused-qkv splits / nope-rope slices read in place
  torch::Tensor xv = x;
  long rstride = (long)nH * D, hstride = D;
  const bool strided_ok = x.stride(3) == 1
      && x.stride(0) == x.size(1) * x.stride(1);
  if (strided_ok) {
    rstride = x.stride(1);
    hstride = x.stride(2);
  } else {
    xv = x.contiguous();
  }
  auto y = torch::empty({B, T, nH, D}, x.options());
  launch_rope(xv.data_ptr(), y.data_ptr(), cc.data_ptr<float>(),
              sc.data_ptr<float>(), (long)B * T, T, nH, D, interleaved,
              rstride, hstride, cur_stream());
  return y;
}

// k/v [B, T, Hkv, D]; caches full buffers [B, Hkv, Scap, D]
void rope_append_kv(torch::Tensor k, torch::Tensor v, torch::Tensor cos,
                    torch::Tensor sin, torch::Tensor kcache,
                    torch::Tensor vcache, c10::optional<torch::Tensor> pos,
                    int64_t pos0, bool interleaved) {
  check_bf16(k, "k");
  TORCH_CHECK(k.dim() == 4, "k must be [B, T, Hkv, D]");
  auto cc = cos.contiguous();
  auto sc = sin.contiguous();
  TORCH_CHECK(cc.scalar_type() == torch::kFloat32, "cos must be fp32");
  const int B = k.size(0), T = k.size(1), Hkv = k.size(2), D = k.size(3);
  TORCH_CHECK(D % 2 == 0, "head dim must be even");
  TORCH_CHECK(cc.size(0) == T && cc.size(1) == D / 2, "cos shape mismatch");
  TORCH_CHECK(kcache.is_contiguous() && vcache.is_contiguous(),
              "caches contiguous");
  TORCH_CHECK(kcache.size(3) == D && vcache.size(3) == D, "cache D mismatch");
  const long Scap = kcache.size(2);
  const int* pp = nullptr;
  if (pos.has_value()) pp = pos->data_ptr<int>();
  // fused-qkv split views read in place (heads contiguous within slice)
  auto view_ok = [&](const torch::Tensor& t) {
    return t.stride(3) == 1 && t.stride(2) == D
        && t.stride(0) == t.size(1) * t.stride(1);
  };
  torch::Tensor kv = k, vv = v;
  long krs = (long)Hkv * D, vrs = (long)Hkv * D;
  if (view_ok(k)) krs = k.stride(1); else kv = k.contiguous();
  if (view_ok(v)) vrs = v.stride(1); else vv = v.contiguous();
  launch_rope_append_kv(kv.data_ptr(), vv.data_ptr(), cc.data_ptr<float>(),
                        sc.data_ptr<float>(), kcache.data_ptr(),
                        vcache.data_ptr(), pp, (int)pos0, B, T, Hkv, D, Scap,
                        interleaved, krs, vrs, cur_stream());
}

// kvh [B, T, nh, nope+vd]; kpe [B, T, rope]; caches full buffers
void mla_append_kv(torch::Tensor kvh, torch::Tensor kpe, torch::Tensor kcache,
                   torch::Tensor vcache, c10::optional<torch::Tensor> pos,
                   int64_t pos0) {
  check_bf16(kvh, "kvh");
  const int B = kvh.size(0), T = kvh.size(1), nh = kvh.size(2);
  const int rope = kpe.size(-1);
  const int kd = kcache.size(3);
  const int vd = vcache.size(3);
  const int nope = kd - rope;
  TORCH_CHECK(kvh.size(3) == nope + vd, "kvh dim mismatch");
  TORCH_CHECK(kcache.is_contiguous() && vcache.is_contiguous(), "caches contiguous");
  const long Scap = kcache.size(2);
  const int* pp = nullptr;
  if (pos.has_value()) pp = pos->data_ptr<int>();
  long kpe_rows, kpe_rs;
  torch::Tensor kpev = kpe;
  if (!row_view2(kpe, kpe_rows, kpe_rs)) {
    kpev = kpe.contiguous();
    kpe_rs = rope;
  }
  launch_mla_append_kv(kvh.contiguous().data_ptr(), kpev.data_ptr(),
                       kcache.data_ptr(), vcache.data_ptr(), pp, (int)pos0, B,
                       T, nh, nope, vd, rope, Scap, kpe_rs, cur_stream());
}

// q [B, Hq, 1, Dk]; k/v: cache views [B, Hkv, S, D] with row-contiguous
// last dim over a [B, Hkv, Scap, D] buffer.
torch::Tensor attn_decode(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                          double scale, double softcap, int64_t window,
                          c10::optional<torch::Tensor> pos) {
  check_bf16(q, "q");
  check_bf16(k, "k");
  TORCH_CHECK(q.size(2) == 1, "attn_decode needs Tq == 1");
  auto qc = q.contiguous();
  const int B = qc.size(0), Hq = qc.size(1), Dk = qc.size(3);
  const int Hkv = k.size(1), S = k.size(2);
  const int Dv = v.size(3);
  TORCH_CHECK(k.stride(3) == 1 && v.stride(3) == 1, "K/V rows must be contiguous");
  TORCH_CHECK(k.stride(2) == Dk, "K seq stride mismatch");
  // V rows may live inside the K rows (absorbed MLA: V = K[..., :Dv])
  const long vstride = v.stride(2);
  TORCH_CHECK(vstride >= Dv, "V seq stride too small");
  TORCH_CHECK(Hq % Hkv == 0, "Hq must be divisible by Hkv");
  TORCH_CHECK(Dk % 8 == 0, "Dk must be a multiple of 8");
  TORCH_CHECK(attn_decode_supported_shape(Hq / Hkv, Dv),
              "unsupported GQA ratio/Dv ", Hq / Hkv, " ", Dv);
  TORCH_CHECK(Dv <= 512, "Dv too large");
  long kScap = k.stride(1) / Dk;
  long vScap = v.stride(1) / vstride;
  TORCH_CHECK(kScap == vScap, "K/V capacity mismatch");
  auto out = torch::empty({B, Hq, 1, Dv}, qc.options());
  const int* s_ptr = nullptr;
  if (pos.has_value()) {
    TORCH_CHECK(pos->scalar_type() == torch::kInt32, "pos must be int32");
    s_ptr = pos->data_ptr<int>();
  }
  // split-S for parallelism: target ~1024 blocks, slices of >= 512 keys
  // (>= 256 keys when B*Hkv alone leaves the chip block-starved, e.g.
  // absorbed MLA with Hkv = 1)
  long span = s_ptr ? kScap : (long)S;
  long slice = (B * Hkv <= 64) ? 256 : 512;
  int nsplit = (int)std::min<long>(
      std::min<long>(8, std::max<long>(1, 1024 / std::max(1, B * Hkv))),
      std::max<long>(1, (span + slice - 1) / slice));
  float* part_o = nullptr;
  float* part_ml = nullptr;
  torch::Tensor po, pml;
  if (nsplit > 1) {
    auto fopts = qc.options().dtype(torch::kFloat32);
    po = torch::empty({(long)B * Hq * nsplit * Dv}, fopts);
    pml = torch::empty({(long)B * Hq * nsplit * 2}, fopts);
    part_o = po.data_ptr<float>();
    part_ml = pml.data_ptr<float>();
  }
  launch_attn_decode(qc.data_ptr(), k.data_ptr(), v.data_ptr(),
                     out.data_ptr(), part_o, part_ml, nsplit, s_ptr, B, Hq,
                     Hkv, S, kScap, Dk, Dv, vstride, (float)scale,
                     (float)softcap, (int)window, cur_stream());
  return out;
}

// q [B, Hq, T, Dk]; k/v cache views [B, Hkv, S, D] (row-contiguous)
torch::Tensor attn_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                           double scale, double softcap, int64_t window,
                           int64_t causal_offset) {
  check_bf16(q, "q");
  auto qc = q.contiguous();
  const int B = qc.size(0), Hq = qc.size(1), T = qc.size(2), Dk = qc.size(3);
  const int Hkv = k.size(1), S = k.size(2), Dv = v.size(3);
  TORCH_CHECK(k.stride(3) == 1 && v.stride(3) == 1, "K/V rows must be contiguous");
  TORCH_CHECK(k.stride(2) == Dk && v.stride(2) == Dv, "K/V seq stride mismatch");
  TORCH_CHECK(Hq % Hkv == 0 && Dk % 32 == 0 && Dv % 16 == 0, "shape unsupported");
  TORCH_CHECK(Dk <= 256 && Dv <= 256, "Dk<=256, Dv<=256");
  // NEVER silently skip the launch: an uninstantiated template combo
  // once fell through and returned an uninitialized output
  TORCH_CHECK(attn_prefill_supported(Dk, Dv),
              "attn_prefill: no kernel instantiation for Dk=", Dk,
              " Dv=", Dv);
  long kScap = k.stride(1) / Dk;
  long vScap = v.stride(1) / Dv;
  auto out = torch::empty({B, Hq, T, Dv}, qc.options());
  launch_attn_prefill(qc.data_ptr(), k.data_ptr(), v.data_ptr(),
                      out.data_ptr(), B, Hq, Hkv, T, S, kScap, vScap, Dk, Dv,
                      (float)scale, (float)softcap, (int)window,
                      (int)causal_offset, cur_stream());
  return out;
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor Bm) {
  check_bf16(A, "A");
  auto D = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
  launch_mfma_probe(A.contiguous().data_ptr(), Bm.contiguous().data_ptr(),
                    D.data_ptr<float>(), cur_stream());
  return D;
}

// x [M, H] bf16, wq [O, H*bits/32] uint32/int32, scales/biases [O, H/gs]
// Dense bf16 decode GEMV: y = x @ w^T on MFMA (M <= 64)
torch::Tensor dense_gemv(torch::Tensor x, torch::Tensor w) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  auto xc = x.contiguous();
  const int M = xc.size(0), H = xc.size(1);
  const int O = w.size(0);
  TORCH_CHECK(w.is_contiguous(), "w must be contiguous");
  TORCH_CHECK(M <= 64 && H % 32 == 0, "dense_gemv needs M<=64, H%32==0");
  auto y = torch::empty({M, O}, xc.options());
  const int nk = bf16_gemv_nsplit(M, O, H);
  torch::Tensor yf;
  float* yfp = nullptr;
  if (nk > 1) {
    yf = torch::empty({M, O}, xc.options().dtype(torch::kFloat32));
    yfp = yf.data_ptr<float>();
  }
  launch_bf16_gemv_mfma(xc.data_ptr(), w.data_ptr(), y.data_ptr(), yfp, nk,
                        M, O, H, cur_stream());
  return y;
}

// K-segmented coalesced-A dense bf16 GEMM (gemm_kseg.hip): targets the
// K-long down_proj shapes where hipBLASLt plateaus (~3.3 TB/s)
torch::Tensor gemm_m64_kseg(torch::Tensor x, torch::Tensor w,
                            int64_t ksegs) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  auto xc = x.contiguous();
  const int M = xc.size(0), K = xc.size(1);
  const int N = w.size(0);
  TORCH_CHECK(w.is_contiguous(), "w must be contiguous");
  TORCH_CHECK(M >= 1 && M <= 64 && K % 256 == 0,
              "gemm_m64_kseg needs 1<=M<=64, K%256==0");
  TORCH_CHECK(ksegs >= 1);
  auto yf = torch::empty({64, N}, xc.options().dtype(torch::kFloat32));
  auto y = torch::empty({M, N}, xc.options());
  launch_gemm_kseg(xc.data_ptr(), w.data_ptr(), yf.data_ptr<float>(),
                   y.data_ptr(), M, N, K, (int)ksegs, cur_stream());
  return y;
}

// w4a16 K-segmented GEMM: packed repacked weights (ops.repack_w4) +
// fp16 activations; bf16 out
torch::Tensor gemm_m64_kseg_w4(torch::Tensor x, torch::Tensor wq,
                               torch::Tensor sc, torch::Tensor bi,
                               int64_t gs, int64_t ksegs) {
  TORCH_CHECK(x.scalar_type() == torch::kHalf, "x must be fp16");
  auto xc = x.contiguous();
  const int M = xc.size(0), K = xc.size(1);
  const int N = wq.size(0);
  TORCH_CHECK(wq.is_contiguous() && sc.is_contiguous() && bi.is_contiguous());
  TORCH_CHECK(M >= 1 && M <= 64 && K % 256 == 0,
              "gemm_m64_kseg_w4 needs 1<=M<=64, K%256==0");
  TORCH_CHECK(gs == 32 || gs == 64 || gs == 128, "gs must be 32/64/128");
  TORCH_CHECK(wq.size(1) == K / 8, "wq must be repacked w4 [N, K/8]");
  auto yf = torch::empty({64, N}, xc.options().dtype(torch::kFloat32));
  auto y = torch::empty({M, N}, xc.options().dtype(torch::kBFloat16));
  launch_gemm_kseg_w4(xc.data_ptr(), wq.data_ptr(), sc.data_ptr(),
                      bi.data_ptr(), yf.data_ptr<float>(), y.data_ptr(), M, N,
                      K, (int)gs, (int)ksegs, cur_stream());
  return y;
}

// LDS-tiled dense bf16 GEMM, M <= 64, deep-k shapes
torch::Tensor dense_gemm64(torch::Tensor x, torch::Tensor w) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  auto xc = x.contiguous();
  const int M = xc.size(0), H = xc.size(1);
  const int O = w.size(0);
  TORCH_CHECK(w.is_contiguous(), "w must be contiguous");
  TORCH_CHECK(M <= 64 && H % 8 == 0, "dense_gemm64 needs M<=64, H%8==0");
  auto y = torch::empty({M, O}, xc.options());
  const int nk = bf16_gemm_m64_nsplit(M, O, H);
  torch::Tensor yf;
  float* yfp = nullptr;
  if (nk > 1) {
    yf = torch::empty({M, O}, xc.options().dtype(torch::kFloat32));
    yfp = yf.data_ptr<float>();
  }
  launch_bf16_gemm_m64(xc.data_ptr(), w.data_ptr(), y.data_ptr(), yfp, nk, M,
                       O, H, cur_stream());
  return y;
}

torch::Tensor w4a16_gemv(torch::Tensor x, torch::Tensor wq,
                         torch::Tensor scales, torch::Tensor biases,
                         int64_t gs, int64_t bits) {
  check_bf16(x, "x");
  auto xc = x.contiguous();
  const int M = xc.size(0), H = xc.size(1);
  const int O = wq.size(0);
  auto y = torch::empty({M, O}, xc.options());
  if (M >= 8 && H % 32 == 0 && gs % 32 == 0) {
    const int nk = w4a16_mfma_nsplit(M, O, H);
    torch::Tensor yf;
    float* yfp = nullptr;
    if (nk > 1) {
      yf = torch::empty({M, O}, xc.options().dtype(torch::kFloat32));
      yfp = yf.data_ptr<float>();
    }
    launch_w4a16_mfma(xc.data_ptr(), wq.contiguous().data_ptr(),
                      scales.contiguous().data_ptr(),
                      biases.contiguous().data_ptr(), y.data_ptr(), yfp, nk,
                      M, O, H, (int)gs, (int)bits, cur_stream());
  } else {
    launch_w4a16_gemv(xc.data_ptr(), wq.contiguous().data_ptr(),
                      scales.contiguous().data_ptr(),
                      biases.contiguous().data_ptr(), y.data_ptr(), M, O, H,
                      (int)gs, (int)bits, cur_stream());
  }
  return y;
}

torch::Tensor dequant(torch::Tensor wq, torch::Tensor scales,
                      torch::Tensor biases, int64_t H, int64_t gs,
                      int64_t bits) {
  const long O = wq.size(0);
  auto out = torch::empty({O, H}, scales.options());
  launch_dequant(wq.contiguous().data_ptr(), scales.contiguous().data_ptr(),
                 biases.contiguous().data_ptr(), out.data_ptr(), O, (int)H,
                 (int)gs, (int)bits, cur_stream());
  return out;
}

torch::Tensor moe_gateup_grouped(torch::Tensor x, torch::Tensor gate_w,
                                 torch::Tensor up_w, torch::Tensor sub_expert,
                                 torch::Tensor sub_off, torch::Tensor sub_cnt,
                                 torch::Tensor sorted_tok, int64_t P,
                                 int64_t max_tok) {
  check_bf16(x, "x");
  const int H = x.size(1);
  const int I = gate_w.size(1);
  const int S = sub_expert.size(0);
  TORCH_CHECK(sub_expert.scalar_type() == torch::kInt32, "sub arrays int32");
  auto h = torch::empty({P, I}, x.options());
  if (max_tok == 16 && H % 32 == 0 && I % 16 == 0) {
    launch_moe_gateup_mfma(
        x.contiguous().data_ptr(), gate_w.data_ptr(), up_w.data_ptr(),
        h.data_ptr(), sub_expert.data_ptr<int>(), sub_off.data_ptr<int>(),
        sub_cnt.data_ptr<int>(), sorted_tok.data_ptr<int>(), S, H, I,
        cur_stream());
  } else {
    TORCH_CHECK(max_tok <= 4, "scalar grouped kernel needs max_tok <= 4");
    launch_moe_gateup_grouped(
        x.contiguous().data_ptr(), gate_w.data_ptr(), up_w.data_ptr(),
        h.data_ptr(), sub_expert.data_ptr<int>(), sub_off.data_ptr<int>(),
        sub_cnt.data_ptr<int>(), sorted_tok.data_ptr<int>(), S, H, I,
        cur_stream());
  }
  return h;
}

torch::Tensor moe_down_grouped(torch::Tensor h, torch::Tensor down_w,
                               torch::Tensor sub_expert, torch::Tensor sub_off,
                               torch::Tensor sub_cnt, torch::Tensor sorted_tok,
                               torch::Tensor sorted_wt, int64_t N,
                               int64_t max_tok) {
  const int I = h.size(1);
  const int H = down_w.size(1);
  const int S = sub_expert.size(0);
  auto out = torch::zeros({N, H}, h.options().dtype(torch::kFloat32));
  if (max_tok == 16 && I % 32 == 0 && H % 16 == 0) {
    launch_moe_down_mfma(
        h.contiguous().data_ptr(), down_w.data_ptr(), out.data_ptr<float>(),
        sub_expert.data_ptr<int>(), sub_off.data_ptr<int>(),
        sub_cnt.data_ptr<int>(), sorted_tok.data_ptr<int>(),
        sorted_wt.data_ptr<float>(), S, I, H, cur_stream());
  } else {
    TORCH_CHECK(max_tok <= 4, "scalar grouped kernel needs max_tok <= 4");
    launch_moe_down_grouped(
        h.contiguous().data_ptr(), down_w.data_ptr(), out.data_ptr<float>(),
        sub_expert.data_ptr<int>(), sub_off.data_ptr<int>(),
        sub_cnt.data_ptr<int>(), sorted_tok.data_ptr<int>(),
        sorted_wt.data_ptr<float>(), S, I, H, cur_stream());
  }
  return out;
}

torch::Tensor moe_w4_mfma(torch::Tensor x, torch::Tensor wq,
                          torch::Tensor scales, torch::Tensor biases,
                          torch::Tensor sub_expert, torch::Tensor sub_off,
                          torch::Tensor sub_cnt, torch::Tensor sorted_tok,
                          int64_t P, int64_t gs, int64_t bits) {
  check_bf16(x, "x");
  const int H = x.size(1);
  const int O = wq.size(1);
  const int S = sub_expert.size(0);
  TORCH_CHECK(H % 32 == 0 && gs % 32 == 0, "H%32, gs%32 required");
  auto y = torch::empty({P, O}, x.options());
  launch_moe_w4_mfma(x.contiguous().data_ptr(), wq.data_ptr(),
                     scales.data_ptr(), biases.data_ptr(), y.data_ptr(),
                     sub_expert.data_ptr<int>(), sub_off.data_ptr<int>(),
                     sub_cnt.data_ptr<int>(), sorted_tok.data_ptr<int>(), S,
                     H, O, (int)gs, (int)bits, cur_stream());
  return y;
}

// Fused fp16-dequant gate+up+SiLU over repacked quant words.
// x [N, H] fp16; gq/uq [E, I, H*bits/32] repacked; scales/biases bf16.
torch::Tensor moe_w4f16_gateup(torch::Tensor x, torch::Tensor gq,
                               torch::Tensor uq, torch::Tensor gsc,
                               torch::Tensor gbi, torch::Tensor usc,
                               torch::Tensor ubi, torch::Tensor sub_expert,
                               torch::Tensor sub_off, torch::Tensor sub_cnt,
                               torch::Tensor sorted_tok, int64_t P,
                               int64_t gs, int64_t bits) {
  TORCH_CHECK(x.scalar_type() == torch::kHalf, "x must be fp16");
  const int H = x.size(1);
  const int I = gq.size(1);
  const int S = sub_expert.size(0);
  TORCH_CHECK(H % 32 == 0, "H%32 required");
  TORCH_CHECK(gs == 32 || gs == 64 || gs == 128, "gs must be 32/64/128");
  auto h = torch::empty({P, I}, x.options());
  launch_moe_w4f16_gateup(
      x.contiguous().data_ptr(), gq.data_ptr(), uq.data_ptr(), gsc.data_ptr(),
      gbi.data_ptr(), usc.data_ptr(), ubi.data_ptr(), h.data_ptr(),
      sub_expert.data_ptr<int>(), sub_off.data_ptr<int>(),
      sub_cnt.data_ptr<int>(), sorted_tok.data_ptr<int>(), S, H, I, (int)gs,
      (int)bits, cur_stream());
  return h;
}

torch::Tensor moe_w4f16_down(torch::Tensor hh, torch::Tensor dq,
                             torch::Tensor dsc, torch::Tensor dbi,
                             torch::Tensor sub_expert, torch::Tensor sub_off,
                             torch::Tensor sub_cnt, torch::Tensor sorted_tok,
                             torch::Tensor sorted_wt, int64_t N, int64_t gs,
                             int64_t bits) {
  TORCH_CHECK(hh.scalar_type() == torch::kHalf, "h must be fp16");
  const int I = hh.size(1);
  const int H = dq.size(1);
  const int S = sub_expert.size(0);
  TORCH_CHECK(I % 32 == 0, "I%32 required");
  TORCH_CHECK(gs == 32 || gs == 64 || gs == 128, "gs must be 32/64/128");
  auto out = torch::zeros({N, H}, hh.options().dtype(torch::kFloat32));
  launch_moe_w4f16_down(
      hh.contiguous().data_ptr(), dq.data_ptr(), dsc.data_ptr(),
      dbi.data_ptr(), out.data_ptr<float>(), sub_expert.data_ptr<int>(),
      sub_off.data_ptr<int>(), sub_cnt.data_ptr<int>(),
      sorted_tok.data_ptr<int>(), sorted_wt.data_ptr<float>(), S, I, H,
      (int)gs, (int)bits, cur_stream());
  return out;
}

// Dense fp16-dequant w4/w8 GEMV: x [M, H] fp16, wq REPACKED
// (ops.repack_w4), scales/biases bf16.  Returns bf16 [M, O].
torch::Tensor w4f16_gemv(torch::Tensor x, torch::Tensor wq,
                         torch::Tensor scales, torch::Tensor biases,
                         int64_t gs, int64_t bits) {
  TORCH_CHECK(x.scalar_type() == torch::kHalf, "x must be fp16");
  auto xc = x.contiguous();
  const int M = xc.size(0), H = xc.size(1);
  const int O = wq.size(0);
  TORCH_CHECK(M <= 64 && H % 32 == 0, "w4f16_gemv needs M<=64, H%32==0");
  TORCH_CHECK(gs == 32 || gs == 64 || gs == 128, "gs must be 32/64/128");
  auto y = torch::empty({M, O},
                        xc.options().dtype(torch::kBFloat16));
  const int nk = w4f16_gemv_nsplit(M, O, H);
  torch::Tensor yf;
  float* yfp = nullptr;
  if (nk > 1) {
    yf = torch::empty({M, O}, xc.options().dtype(torch::kFloat32));
    yfp = yf.data_ptr<float>();
  }
  launch_w4f16_gemv(xc.data_ptr(), wq.contiguous().data_ptr(),
                    scales.contiguous().data_ptr(),
                    biases.contiguous().data_ptr(), y.data_ptr(), yfp, nk, M,
                    O, H, (int)gs, (int)bits, cur_stream());
  return y;
}

std::vector<torch::Tensor> moe_gate_subranges(torch::Tensor logits, int64_t K,
                                              int64_t s_upper, int64_t max_tok,
                                              double routed_scaling,
                                              bool norm_topk) {
  check_bf16(logits, "logits");
  auto lc = logits.contiguous();
  const int N = lc.size(0), E = lc.size(1);
  TORCH_CHECK(N <= 128 && E <= 64 && K <= 8,
              "fused gate limits: N<=128, E<=64, K<=8");
  const long P = (long)N * K;
  auto opts = torch::TensorOptions().dtype(torch::kInt32).device(lc.device());
  auto sorted_tok = torch::empty({P}, opts);
  auto sorted_wt = torch::empty({P}, opts.dtype(torch::kFloat32));
  auto sub_expert = torch::empty({s_upper}, opts);
  auto sub_off = torch::empty({s_upper}, opts);
  auto sub_cnt = torch::empty({s_upper}, opts);
  launch_moe_gate_subranges(lc.data_ptr(), sorted_tok.data_ptr<int>(),
                            sorted_wt.data_ptr<float>(),
                            sub_expert.data_ptr<int>(), sub_off.data_ptr<int>(),
                            sub_cnt.data_ptr<int>(), N, E, (int)K,
                            (int)s_upper, (int)max_tok, (float)routed_scaling,
                            norm_topk ? 1 : 0, cur_stream());
  return {sub_expert, sub_off, sub_cnt, sorted_tok, sorted_wt};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rms_norm", &rms_norm, "fused RMSNorm (gfx950)");
  m.def("rms_norm_residual", &rms_norm_residual);
  m.def("glu", &glu);
  m.def("softcap", &softcap_op);
  m.def("apply_rope", &apply_rope);
  m.def("rope_append_kv", &rope_append_kv, pybind11::arg("k"),
        pybind11::arg("v"), pybind11::arg("cos"), pybind11::arg("sin"),
        pybind11::arg("kcache"), pybind11::arg("vcache"),
        pybind11::arg("pos") = pybind11::none(), pybind11::arg("pos0") = 0,
        pybind11::arg("interleaved") = false);
  m.def("mla_append_kv", &mla_append_kv,
        pybind11::arg("kvh"), pybind11::arg("kpe"), pybind11::arg("kcache"),
        pybind11::arg("vcache"), pybind11::arg("pos") = pybind11::none(),
        pybind11::arg("pos0") = 0);
  m.def("attn_decode", &attn_decode, pybind11::arg("q"), pybind11::arg("k"),
        pybind11::arg("v"), pybind11::arg("scale"), pybind11::arg("softcap"),
        pybind11::arg("window"), pybind11::arg("pos") = pybind11::none());
  m.def("attn_decode_shape_ok", [](int64_t g, int64_t dv) {
    return attn_decode_supported_shape((int)g, (int)dv);
  });
  m.def("attn_prefill", &attn_prefill);
  m.def("attn_prefill_shape_ok", [](int64_t dk, int64_t dv) {
    return attn_prefill_supported((int)dk, (int)dv);
  });
  m.def("mfma_probe", &mfma_probe);
  m.def("w4a16_gemv", &w4a16_gemv);
  m.def("w4f16_gemv", &w4f16_gemv);
  m.def("dense_gemv", &dense_gemv);
  m.def("dense_gemm64", &dense_gemm64);
  m.def("gemm_m64_kseg", &gemm_m64_kseg);
  m.def("gemm_m64_kseg_w4", &gemm_m64_kseg_w4);
  m.def("dequant", &dequant);
  m.def("moe_gateup_grouped", &moe_gateup_grouped);
  m.def("moe_down_grouped", &moe_down_grouped);
  m.def("moe_w4_mfma", &moe_w4_mfma);
  m.def("moe_w4f16_gateup", &moe_w4f16_gateup);
  m.def("moe_w4f16_down", &moe_w4f16_down);
  m.def("moe_gate_subranges", &moe_gate_subranges);
  m.def("moe_scatter_rows",
        [](torch::Tensor src, torch::Tensor dst, torch::Tensor src_idx,
           torch::Tensor dst_idx) {
          check_bf16(src, "src");
          const int P = src_idx.size(0), H = src.size(-1);
          TORCH_CHECK(H % 4 == 0 && dst.size(-1) == H, "row shape");
          TORCH_CHECK(src_idx.scalar_type() == torch::kInt32 &&
                          dst_idx.scalar_type() == torch::kInt32,
                      "indices int32");
          launch_moe_scatter_rows(src.contiguous().data_ptr(), dst.data_ptr(),
                                  src_idx.data_ptr<int>(),
                                  dst_idx.data_ptr<int>(), P, H,
                                  cur_stream());
        });
  m.def("moe_gather_reduce",
        [](torch::Tensor d, torch::Tensor pos, torch::Tensor wts, int64_t N) {
          check_bf16(d, "d");
          const int K = pos.size(1), H = d.size(-1);
          TORCH_CHECK(pos.scalar_type() == torch::kInt32, "pos int32");
          TORCH_CHECK(wts.scalar_type() == torch::kFloat32, "wts fp32");
          auto out = torch::empty({N, H}, d.options());
          launch_moe_gather_reduce(d.contiguous().data_ptr(),
                                   pos.contiguous().data_ptr<int>(),
                                   wts.contiguous().data_ptr<float>(),
                                   out.data_ptr(), (int)N, K, H,
                                   cur_stream());
          return out;
        });
}
