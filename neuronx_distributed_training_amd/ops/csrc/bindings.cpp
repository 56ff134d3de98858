// PyTorch bindings for the MI355X HIP kernels (gfx950-only build).
#include <torch/extension.h>

#include <hip/hip_runtime.h>

#include <c10/hip/HIPStream.h>

extern "C" {
void launch_rmsnorm_fwd(const void*, const void*, void*, void*, long, int,
                        float, hipStream_t);
void launch_rmsnorm_bwd(const void*, const void*, const void*, const void*,
                        void*, void*, long, int, hipStream_t);
void launch_swiglu_fwd(const void*, void*, long, int, hipStream_t);
void launch_swiglu_bwd(const void*, const void*, void*, long, int,
                       hipStream_t);
void launch_rope_fwd(const void*, const void*, const void*, void*, int, int,
                     int, int, const long*, int, int, float, hipStream_t);
void launch_adamw(void*, const void*, void*, void*, const void*, const void*,
                  void*, long, float, float, float, float, float, int,
                  hipStream_t);
void launch_flash_fwd(const void*, const void*, const void*, void*, void*,
                      int, int, int, int, int, int, bool, float, int,
                      const long*, const long*, const long*, hipStream_t);
void launch_flash_fwd_dbuf(const void*, const void*, const void*, void*,
                           void*, int, int, int, int, int, int, bool, float,
                           int, const long*, const long*, const long*,
                           hipStream_t);
void launch_flash_bwd(const void*, const void*, const void*, const void*,
                      const void*, const void*, void*, void*, void*, int, int,
                      int, int, int, int, bool, float, int, const long*,
                      const long*, const long*, const long*, hipStream_t);
void launch_attn_delta(const void*, const void*, void*, int, int, int, int,
                       const long*, const long*, hipStream_t);
void launch_mfma_probe(const void*, const void*, void*, hipStream_t);
void launch_mfma_probe32(const void*, const void*, void*, hipStream_t);
void launch_flash_fwd_v3(const void*, const void*, const void*, void*, void*,
                         int, int, int, int, int, bool, float, int,
                         const long*, const long*, const long*, hipStream_t);
void launch_flash_bwd_v3(const void*, const void*, const void*, const void*,
                         const void*, const void*, void*, void*, void*, int,
                         int, int, int, int, bool, float, int, const long*,
                         const long*, const long*, const long*, hipStream_t);
void launch_moe_gemm(const void*, const void*, void*, const int*, const int*,
                     int, int, int, long, bool, hipStream_t);
void launch_moe_wgrad(const void*, const void*, void*, const int*, int, int,
                      int, hipStream_t);
void launch_ce_fwd(const void*, const void*, void*, void*, void*, long, int,
                   long, hipStream_t);
void launch_ce_bwd(const void*, const void*, const void*, const void*,
                   const void*, void*, long, int, long, hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_IN(t)                                             \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");             \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  CHECK_IN(x);
  CHECK_IN(w);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "rmsnorm: bf16 only");
  auto N = x.size(0);
  auto H = x.size(1);
  TORCH_CHECK(H % 8 == 0, "hidden must be divisible by 8");
  auto y = torch::empty_like(x);
  auto invrms = torch::empty({N}, x.options().dtype(torch::kFloat32));
  launch_rmsnorm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                     invrms.data_ptr(), N, (int)H, (float)eps, cur_stream());
  return {y, invrms};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor invrms) {
  CHECK_IN(dy);
  CHECK_IN(x);
  auto N = x.size(0);
  auto H = x.size(1);
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({H}, x.options().dtype(torch::kFloat32));
  launch_rmsnorm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                     invrms.data_ptr(), dx.data_ptr(), dw32.data_ptr(), N,
                     (int)H, cur_stream());
  return {dx, dw32.to(w.scalar_type())};
}

torch::Tensor swiglu_fwd(torch::Tensor gu) {
  CHECK_IN(gu);
  auto N = gu.size(0);
  auto I = gu.size(1) / 2;
  TORCH_CHECK(I % 8 == 0, "intermediate must be divisible by 8");
  auto y = torch::empty({N, I}, gu.options());
  launch_swiglu_fwd(gu.data_ptr(), y.data_ptr(), N, (int)I, cur_stream());
  return y;
}

torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor gu) {
  CHECK_IN(dy);
  CHECK_IN(gu);
  auto N = gu.size(0);
  auto I = gu.size(1) / 2;
  auto dgu = torch::empty_like(gu);
  launch_swiglu_bwd(dy.data_ptr(), gu.data_ptr(), dgu.data_ptr(), N, (int)I,
                    cur_stream());
  return dgu;
}

torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cost, torch::Tensor sint,
                       long pos_offset, long pos_offset2) {
  // x: logical [b, h, s, d], any strides with d contiguous (e.g. a view of
  // the fused QKV output). Returns a [s, b, h, d]-contiguous tensor exposed
  // as a [b, h, s, d] view — zero permute copies in the attention path.
  // pos_offset2 >= 0: zigzag CP layout — rows [s/2, s) take positions
  // pos_offset2 + i instead of pos_offset + s/2 + i.
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.stride(3) == 1,
              "rope: x must be [b, h, s, d] on GPU with d contiguous");
  auto cos_c = cost.contiguous();
  auto sin_c = sint.contiguous();
  TORCH_CHECK(cos_c.scalar_type() == torch::kFloat32, "rope table must be f32");
  int B = (int)x.size(0), H = (int)x.size(1), S = (int)x.size(2),
      D = (int)x.size(3);
  TORCH_CHECK(pos_offset2 < 0 || S % 2 == 0, "zigzag rope needs even s");
  auto y_mem = torch::empty({S, B, H, D}, x.options());
  long xstr[3] = {x.stride(2), x.stride(0), x.stride(1)};  // s, b, h
  launch_rope_fwd(x.data_ptr(), cos_c.data_ptr(), sin_c.data_ptr(),
                  y_mem.data_ptr(), S, B, H, D, xstr, (int)pos_offset,
                  (int)pos_offset2, 1.0f, cur_stream());
  return y_mem.permute({1, 2, 0, 3});
}

void adamw_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, torch::Tensor wd_mask, double lr, double b1,
                double b2, double eps, double wd, long step,
                c10::optional<torch::Tensor> grad_scale,
                c10::optional<torch::Tensor> p_bf16) {
  CHECK_IN(p);
  CHECK_IN(g);
  void* gs = nullptr;
  if (grad_scale.has_value()) {
    TORCH_CHECK(grad_scale->scalar_type() == torch::kFloat32 &&
                    grad_scale->is_cuda(),
                "grad_scale must be a device fp32 scalar");
    gs = grad_scale->data_ptr();
  }
  void* pb = nullptr;
  if (p_bf16.has_value()) {
    TORCH_CHECK(p_bf16->scalar_type() == torch::kBFloat16 &&
                    p_bf16->is_contiguous() && p_bf16->numel() == p.numel(),
                "p_bf16 must be contiguous bf16 of the same numel");
    pb = p_bf16->data_ptr();
  }
  launch_adamw(p.data_ptr(), g.data_ptr(), m.data_ptr(), v.data_ptr(),
               wd_mask.data_ptr(), gs, pb, p.numel(), (float)lr, (float)b1,
               (float)b2, (float)eps, (float)wd, (int)step, cur_stream());
}

static void check_bhsd(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.dim() == 4 && t.stride(3) == 1,
              name, " must be [b, h, s, d] on GPU with d contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
}

static torch::Tensor attn_delta(torch::Tensor dout, torch::Tensor o) {
  // fused rowsum(dO . O) -> [B, HQ, SQ] fp32 (no fp32 materialization)
  auto oc = o.stride(3) == 1 ? o : o.contiguous();
  int B = (int)dout.size(0), HQ = (int)dout.size(1), SQ = (int)dout.size(2),
      D = (int)dout.size(3);
  auto delta = torch::empty(
      {B, HQ, SQ}, dout.options().dtype(torch::kFloat32));
  long ds[3] = {dout.stride(2), dout.stride(0), dout.stride(1)};
  long os_[3] = {oc.stride(2), oc.stride(0), oc.stride(1)};
  launch_attn_delta(dout.data_ptr(), oc.data_ptr(), delta.data_ptr(), SQ, B,
                    HQ, D, ds, os_, cur_stream());
  return delta;
}

std::vector<torch::Tensor> flash_attn_fwd_v3(torch::Tensor q, torch::Tensor k,
                                             torch::Tensor v, bool causal,
                                             double scale, long window) {
  // T12 swapped-operand forward (ROADMAP §1); dark until HW-validated
  check_bhsd(q, "q");
  check_bhsd(k, "k");
  check_bhsd(v, "v");
  int B = (int)q.size(0), HQ = (int)q.size(1), S = (int)q.size(2),
      D = (int)q.size(3);
  int HKV = (int)k.size(1);
  TORCH_CHECK(D == 128 || D == 64, "head_dim must be 64 or 128");
  TORCH_CHECK(k.size(2) == S, "v3 kernel requires S_q == S_kv (use v2)");
  auto o_mem = torch::empty({S, B, HQ, D}, q.options());
  auto lse = torch::empty({B, HQ, S}, q.options().dtype(torch::kFloat32));
  long qs[3] = {q.stride(2), q.stride(0), q.stride(1)};
  long ks[3] = {k.stride(2), k.stride(0), k.stride(1)};
  long vs[3] = {v.stride(2), v.stride(0), v.stride(1)};
  launch_flash_fwd_v3(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                      o_mem.data_ptr(), lse.data_ptr(), B, HQ, HKV, S, D,
                      causal, (float)scale, (int)window, qs, ks, vs,
                      cur_stream());
  return {o_mem.permute({1, 2, 0, 3}), lse};
}

std::vector<torch::Tensor> flash_attn_bwd_v3(torch::Tensor dout,
                                             torch::Tensor q, torch::Tensor k,
                                             torch::Tensor v, torch::Tensor o,
                                             torch::Tensor lse, bool causal,
                                             double scale, long window) {
  if (dout.stride(3) != 1) dout = dout.contiguous();
  check_bhsd(dout, "dout");
  check_bhsd(q, "q");
  int B = (int)q.size(0), HQ = (int)q.size(1), S = (int)q.size(2),
      D = (int)q.size(3);
  int HKV = (int)k.size(1);
  auto delta = attn_delta(dout, o);
  auto dq_mem = torch::empty({S, B, HQ, D}, q.options());
  int group = HQ / HKV;
  auto f32 = q.options().dtype(torch::kFloat32);
  auto dk_part = torch::empty({group, S, B, HKV, D}, f32);
  auto dv_part = torch::empty({group, S, B, HKV, D}, f32);
  long qs[3] = {q.stride(2), q.stride(0), q.stride(1)};
  long ks[3] = {k.stride(2), k.stride(0), k.stride(1)};
  long vs[3] = {v.stride(2), v.stride(0), v.stride(1)};
  long dstr[3] = {dout.stride(2), dout.stride(0), dout.stride(1)};
  launch_flash_bwd_v3(dout.data_ptr(), q.data_ptr(), k.data_ptr(),
                      v.data_ptr(), lse.data_ptr(), delta.data_ptr(),
                      dq_mem.data_ptr(), dk_part.data_ptr(),
                      dv_part.data_ptr(), B, HQ, HKV, S, D, causal,
                      (float)scale, (int)window, qs, ks, vs, dstr,
                      cur_stream());
  auto dk_mem = dk_part.sum(0).to(torch::kBFloat16);
  auto dv_mem = dv_part.sum(0).to(torch::kBFloat16);
  return {dq_mem.permute({1, 2, 0, 3}), dk_mem.permute({1, 2, 0, 3}),
          dv_mem.permute({1, 2, 0, 3})};
}

std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, bool causal,
                                          double scale, long window) {
  // q/k/v: logical [b, h, s, d], arbitrary strides (views of the QKV GEMM
  // output). Returns O as a [b, h, s, d] view of [s, b, h, d] storage.
  // S_q != S_kv supported; causal is then bottom-right aligned (decode /
  // ring-attention half-block convention, requires S_kv >= S_q).
  check_bhsd(q, "q");
  check_bhsd(k, "k");
  check_bhsd(v, "v");
  int B = (int)q.size(0), HQ = (int)q.size(1), SQ = (int)q.size(2),
      D = (int)q.size(3);
  int HKV = (int)k.size(1), SKV = (int)k.size(2);
  TORCH_CHECK(D == 128 || D == 64, "head_dim must be 64 or 128");
  TORCH_CHECK(HQ % HKV == 0, "GQA head mismatch");
  TORCH_CHECK(!causal || SKV >= SQ, "causal needs S_kv >= S_q");
  auto o_mem = torch::empty({SQ, B, HQ, D}, q.options());
  auto lse = torch::empty({B, HQ, SQ}, q.options().dtype(torch::kFloat32));
  long qs[3] = {q.stride(2), q.stride(0), q.stride(1)};
  long ks[3] = {k.stride(2), k.stride(0), k.stride(1)};
  long vs[3] = {v.stride(2), v.stride(0), v.stride(1)};
  // double-buffered staging variant: bit-identical output, measured
  // +3.0% (bench shape) / +1.8% (TP=8 shape) over the single-buffer
  // kernel -> default (the single-buffer launcher remains for A/B)
  launch_flash_fwd_dbuf(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                        o_mem.data_ptr(), lse.data_ptr(), B, HQ, HKV, SQ,
                        SKV, D, causal, (float)scale, (int)window, qs, ks,
                        vs, cur_stream());
  return {o_mem.permute({1, 2, 0, 3}), lse};
}

std::vector<torch::Tensor> flash_attn_fwd_sbuf(torch::Tensor q,
                                               torch::Tensor k,
                                               torch::Tensor v, bool causal,
                                               double scale, long window) {
  // A/B reference: the SINGLE-buffered staging kernel (two barriers per
  // tile) that the double-buffered default replaced — bit-identical
  check_bhsd(q, "q");
  check_bhsd(k, "k");
  check_bhsd(v, "v");
  int B = (int)q.size(0), HQ = (int)q.size(1), SQ = (int)q.size(2),
      D = (int)q.size(3);
  int HKV = (int)k.size(1), SKV = (int)k.size(2);
  TORCH_CHECK(D == 128 || D == 64, "head_dim must be 64 or 128");
  TORCH_CHECK(!causal || SKV >= SQ, "causal needs S_kv >= S_q");
  auto o_mem = torch::empty({SQ, B, HQ, D}, q.options());
  auto lse = torch::empty({B, HQ, SQ}, q.options().dtype(torch::kFloat32));
  long qs[3] = {q.stride(2), q.stride(0), q.stride(1)};
  long ks[3] = {k.stride(2), k.stride(0), k.stride(1)};
  long vs[3] = {v.stride(2), v.stride(0), v.stride(1)};
  launch_flash_fwd_dbuf(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                        o_mem.data_ptr(), lse.data_ptr(), B, HQ, HKV, SQ,
                        SKV, D, causal, (float)scale, (int)window, qs, ks,
                        vs, cur_stream());
  return {o_mem.permute({1, 2, 0, 3}), lse};
}

std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout, torch::Tensor q,
                                          torch::Tensor k, torch::Tensor v,
                                          torch::Tensor o, torch::Tensor lse,
                                          bool causal, double scale,
                                          long window) {
  if (dout.stride(3) != 1) dout = dout.contiguous();
  check_bhsd(dout, "dout");
  check_bhsd(q, "q");
  int B = (int)q.size(0), HQ = (int)q.size(1), SQ = (int)q.size(2),
      D = (int)q.size(3);
  int HKV = (int)k.size(1), SKV = (int)k.size(2);
  TORCH_CHECK(!causal || SKV >= SQ, "causal needs S_kv >= S_q");
  auto delta = attn_delta(dout, o);  // [B, HQ, SQ] f32, fused
  auto dq_mem = torch::empty({SQ, B, HQ, D}, q.options());
  // dkv writes one fp32 partial slab per q-head of each GQA group (full
  // grid occupancy at high TP); sum + cast here
  int group = HQ / HKV;
  auto f32 = q.options().dtype(torch::kFloat32);
  auto dk_part = torch::empty({group, SKV, B, HKV, D}, f32);
  auto dv_part = torch::empty({group, SKV, B, HKV, D}, f32);
  long qs[3] = {q.stride(2), q.stride(0), q.stride(1)};
  long ks[3] = {k.stride(2), k.stride(0), k.stride(1)};
  long vs[3] = {v.stride(2), v.stride(0), v.stride(1)};
  long ds[3] = {dout.stride(2), dout.stride(0), dout.stride(1)};
  launch_flash_bwd(dout.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                   lse.data_ptr(), delta.data_ptr(), dq_mem.data_ptr(),
                   dk_part.data_ptr(), dv_part.data_ptr(), B, HQ, HKV, SQ,
                   SKV, D, causal, (float)scale, (int)window, qs, ks, vs, ds,
                   cur_stream());
  auto dk_mem = dk_part.sum(0).to(torch::kBFloat16);
  auto dv_mem = dv_part.sum(0).to(torch::kBFloat16);
  return {dq_mem.permute({1, 2, 0, 3}), dk_mem.permute({1, 2, 0, 3}),
          dv_mem.permute({1, 2, 0, 3})};
}

torch::Tensor moe_gemm(torch::Tensor a, torch::Tensor w,
                       torch::Tensor tile_expert, torch::Tensor total_rows,
                       bool trans_b) {
  // a: [Tp, K] bf16 padded/sorted tokens; w: [E, N, Kw] bf16 expert slabs.
  // trans_b=false: C = a @ w[e]^T ([Tp, N]); true: C = a @ w[e] ([Tp, Kw]).
  CHECK_IN(a);
  CHECK_IN(w);
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  w.scalar_type() == torch::kBFloat16,
              "moe_gemm: bf16 only");
  TORCH_CHECK(tile_expert.scalar_type() == torch::kInt &&
                  total_rows.scalar_type() == torch::kInt,
              "moe_gemm: int32 maps");
  long Tp = a.size(0);
  int K = (int)a.size(1);
  int N = (int)w.size(1), Kw = (int)w.size(2);
  TORCH_CHECK(Tp % 256 == 0, "padded rows must be a multiple of 256");
  int row_tiles = (int)(Tp / 256);
  int contraction = trans_b ? N : Kw;
  int out_cols = trans_b ? Kw : N;
  TORCH_CHECK(K == contraction, "moe_gemm: contraction mismatch");
  TORCH_CHECK(out_cols % 128 == 0, "moe_gemm: out cols % 128");
  auto c = torch::empty({Tp, out_cols}, a.options());
  launch_moe_gemm(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                  tile_expert.data_ptr<int>(), total_rows.data_ptr<int>(),
                  row_tiles, K, out_cols, (long)N * Kw, trans_b,
                  cur_stream());
  return c;
}

torch::Tensor moe_wgrad(torch::Tensor dy, torch::Tensor x,
                        torch::Tensor seg_start, long E) {
  // dW[e] = dy[seg_e]^T @ x[seg_e] -> [E, N, K] fp32
  CHECK_IN(dy);
  CHECK_IN(x);
  TORCH_CHECK(seg_start.scalar_type() == torch::kInt, "int32 seg bounds");
  int N = (int)dy.size(1), K = (int)x.size(1);
  TORCH_CHECK(N % 128 == 0 && K % 128 == 0, "moe_wgrad: dims % 128");
  auto dw = torch::empty({E, N, K}, dy.options().dtype(torch::kFloat32));
  launch_moe_wgrad(dy.data_ptr(), x.data_ptr(), dw.data_ptr(),
                   seg_start.data_ptr<int>(), (int)E, N, K, cur_stream());
  return dw;
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b) {
  CHECK_IN(a);
  CHECK_IN(b);
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat32));
  launch_mfma_probe(a.data_ptr(), b.data_ptr(), c.data_ptr(), cur_stream());
  return c;
}

torch::Tensor mfma_probe32(torch::Tensor a, torch::Tensor b) {
  CHECK_IN(a);
  CHECK_IN(b);
  auto c = torch::empty({32, 32}, a.options().dtype(torch::kFloat32));
  launch_mfma_probe32(a.data_ptr(), b.data_ptr(), c.data_ptr(), cur_stream());
  return c;
}

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor targets,
                                  long vocab_start) {
  CHECK_IN(logits);
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16, "ce: bf16 logits");
  TORCH_CHECK(targets.scalar_type() == torch::kLong, "ce: int64 targets");
  long N = logits.size(0);
  int V = (int)logits.size(1);
  auto opt = logits.options().dtype(torch::kFloat32);
  auto row_max = torch::empty({N}, opt);
  auto row_sumexp = torch::empty({N}, opt);
  auto tgt = torch::empty({N}, opt);
  launch_ce_fwd(logits.data_ptr(), targets.contiguous().data_ptr(),
                row_max.data_ptr(), row_sumexp.data_ptr(), tgt.data_ptr(), N,
                V, vocab_start, cur_stream());
  return {row_max, row_sumexp, tgt};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor targets,
                     torch::Tensor gmax, torch::Tensor gsum,
                     torch::Tensor gout, long vocab_start) {
  CHECK_IN(logits);
  long N = logits.size(0);
  int V = (int)logits.size(1);
  auto dl = torch::empty_like(logits);
  launch_ce_bwd(logits.data_ptr(), targets.contiguous().data_ptr(),
                gmax.contiguous().data_ptr(), gsum.contiguous().data_ptr(),
                gout.contiguous().data_ptr(), dl.data_ptr(), N, V,
                vocab_start, cur_stream());
  return dl;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("rope_fwd", &rope_fwd, py::arg("x"), py::arg("cost"), py::arg("sint"),
        py::arg("pos_offset"), py::arg("pos_offset2") = -1);
  m.def("adamw_step", &adamw_step, py::arg("p"), py::arg("g"), py::arg("m"),
        py::arg("v"), py::arg("wd_mask"), py::arg("lr"), py::arg("b1"),
        py::arg("b2"), py::arg("eps"), py::arg("wd"), py::arg("step"),
        py::arg("grad_scale") = py::none(), py::arg("p_bf16") = py::none());
  m.def("flash_attn_fwd_sbuf", &flash_attn_fwd_sbuf, pybind11::arg("q"),
        pybind11::arg("k"), pybind11::arg("v"), pybind11::arg("causal"),
        pybind11::arg("scale"), pybind11::arg("window") = 0);
  m.def("flash_attn_fwd", &flash_attn_fwd, pybind11::arg("q"),
        pybind11::arg("k"), pybind11::arg("v"), pybind11::arg("causal"),
        pybind11::arg("scale"), pybind11::arg("window") = 0);
  m.def("flash_attn_bwd", &flash_attn_bwd, pybind11::arg("dout"),
        pybind11::arg("q"), pybind11::arg("k"), pybind11::arg("v"),
        pybind11::arg("o"), pybind11::arg("lse"), pybind11::arg("causal"),
        pybind11::arg("scale"), pybind11::arg("window") = 0);
  m.def("mfma_probe", &mfma_probe);
  m.def("mfma_probe32", &mfma_probe32);
  m.def("flash_attn_fwd_v3", &flash_attn_fwd_v3, pybind11::arg("q"),
        pybind11::arg("k"), pybind11::arg("v"), pybind11::arg("causal"),
        pybind11::arg("scale"), pybind11::arg("window") = 0);
  m.def("flash_attn_bwd_v3", &flash_attn_bwd_v3, pybind11::arg("dout"),
        pybind11::arg("q"), pybind11::arg("k"), pybind11::arg("v"),
        pybind11::arg("o"), pybind11::arg("lse"), pybind11::arg("causal"),
        pybind11::arg("scale"), pybind11::arg("window") = 0);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("moe_gemm", &moe_gemm);
  m.def("moe_wgrad", &moe_wgrad);
}
