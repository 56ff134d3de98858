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
void launch_rope_fwd(const void*, const void*, const void*, void*, long, int,
                     int, int, float, hipStream_t);
void launch_adamw(void*, const void*, void*, void*, const void*, long, float,
                  float, float, float, float, int, hipStream_t);
void launch_flash_fwd(const void*, const void*, const void*, void*, void*,
                      int, int, int, int, int, bool, float, hipStream_t);
void launch_flash_bwd(const void*, const void*, const void*, const void*,
                      const void*, const void*, void*, void*, void*, int, int,
                      int, int, int, bool, float, hipStream_t);
void launch_mfma_probe(const void*, const void*, void*, hipStream_t);
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
                       long pos_offset) {
  CHECK_IN(x);
  TORCH_CHECK(x.dim() == 4, "rope: x must be [b, h, s, d]");
  auto cos_c = cost.contiguous();
  auto sin_c = sint.contiguous();
  // table may arrive fp32 on CPU buffers moved to GPU; require f32
  TORCH_CHECK(cos_c.scalar_type() == torch::kFloat32, "rope table must be f32");
  long BH = x.size(0) * x.size(1);
  int S = (int)x.size(2);
  int D = (int)x.size(3);
  auto y = torch::empty_like(x);
  float sign = 1.0f;
  launch_rope_fwd(x.data_ptr(), cos_c.data_ptr(), sin_c.data_ptr(),
                  y.data_ptr(), BH, S, D, (int)pos_offset, sign, cur_stream());
  return y;
}

void adamw_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, torch::Tensor wd_mask, double lr, double b1,
                double b2, double eps, double wd, long step) {
  CHECK_IN(p);
  CHECK_IN(g);
  launch_adamw(p.data_ptr(), g.data_ptr(), m.data_ptr(), v.data_ptr(),
               wd_mask.data_ptr(), p.numel(), (float)lr, (float)b1, (float)b2,
               (float)eps, (float)wd, (int)step, cur_stream());
}

std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, bool causal,
                                          double scale) {
  CHECK_IN(q);
  CHECK_IN(k);
  CHECK_IN(v);
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16, "flash_attn: bf16 only");
  int B = (int)q.size(0), HQ = (int)q.size(1), S = (int)q.size(2),
      D = (int)q.size(3);
  int HKV = (int)k.size(1);
  TORCH_CHECK(D == 128 || D == 64, "head_dim must be 64 or 128");
  TORCH_CHECK(HQ % HKV == 0, "GQA head mismatch");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, HQ, S}, q.options().dtype(torch::kFloat32));
  launch_flash_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                   lse.data_ptr(), B, HQ, HKV, S, D, causal, (float)scale,
                   cur_stream());
  return {o, lse};
}

std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout, torch::Tensor q,
                                          torch::Tensor k, torch::Tensor v,
                                          torch::Tensor o, torch::Tensor lse,
                                          bool causal, double scale) {
  CHECK_IN(dout);
  CHECK_IN(q);
  int B = (int)q.size(0), HQ = (int)q.size(1), S = (int)q.size(2),
      D = (int)q.size(3);
  int HKV = (int)k.size(1);
  auto delta = (dout.to(torch::kFloat32) * o.to(torch::kFloat32))
                   .sum(-1)
                   .contiguous();  // [B, HQ, S] f32
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  launch_flash_bwd(dout.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                   lse.data_ptr(), delta.data_ptr(), dq.data_ptr(),
                   dk.data_ptr(), dv.data_ptr(), B, HQ, HKV, S, D, causal,
                   (float)scale, cur_stream());
  return {dq, dk, dv};
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b) {
  CHECK_IN(a);
  CHECK_IN(b);
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat32));
  launch_mfma_probe(a.data_ptr(), b.data_ptr(), c.data_ptr(), cur_stream());
  return c;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("adamw_step", &adamw_step);
  m.def("flash_attn_fwd", &flash_attn_fwd);
  m.def("flash_attn_bwd", &flash_attn_bwd);
  m.def("mfma_probe", &mfma_probe);
}
