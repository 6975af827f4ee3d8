// Python bindings for the edl_amd CDNA4 kernel layer (edl_amd._C).
// Compiled by hipcc directly (no hipify, no CUDA path) — see build_hip.py.
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

// launchers from the .hip translation units
extern "C" void launch_fused_sgd_f32(float*, const float*, float*, float, float,
                                     float, float, long long, hipStream_t);
extern "C" void launch_kd_ce_fwd_f32(const float*, const float*, float*, int, int,
                                     hipStream_t);
extern "C" void launch_kd_ce_bwd_f32(const float*, const float*, float*, float, int,
                                     int, hipStream_t);
extern "C" void launch_kd_ce_fwd_bf16(const void*, const void*, float*, int, int,
                                      hipStream_t);
extern "C" void launch_kd_ce_bwd_bf16(const void*, const void*, void*, float, int,
                                      int, hipStream_t);

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void fused_sgd(torch::Tensor p, torch::Tensor g, torch::Tensor m, double lr,
               double momentum, double weight_decay, double grad_scale) {
  TORCH_CHECK(p.is_cuda() && g.is_cuda() && m.is_cuda(), "fused_sgd: GPU tensors required");
  TORCH_CHECK(p.scalar_type() == torch::kFloat32 && g.scalar_type() == torch::kFloat32 &&
                  m.scalar_type() == torch::kFloat32,
              "fused_sgd: fp32 only");
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous(),
              "fused_sgd: contiguous flat buffers required");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel(), "fused_sgd: size mismatch");
  launch_fused_sgd_f32(p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
                       (float)lr, (float)momentum, (float)weight_decay,
                       (float)grad_scale, (long long)p.numel(), cur_stream());
}

torch::Tensor kd_ce_forward(torch::Tensor s, torch::Tensor t) {
  TORCH_CHECK(s.is_cuda() && t.is_cuda(), "kd_ce: GPU tensors required");
  TORCH_CHECK(s.dim() == 2 && t.sizes() == s.sizes(), "kd_ce: [B,C] logits expected");
  TORCH_CHECK(s.scalar_type() == t.scalar_type(), "kd_ce: dtype mismatch");
  auto sc = s.contiguous();
  auto tc = t.contiguous();
  const int B = (int)s.size(0), C = (int)s.size(1);
  auto lpr = torch::empty({B}, s.options().dtype(torch::kFloat32));
  if (s.scalar_type() == torch::kFloat32) {
    launch_kd_ce_fwd_f32(sc.data_ptr<float>(), tc.data_ptr<float>(),
                         lpr.data_ptr<float>(), B, C, cur_stream());
  } else if (s.scalar_type() == torch::kBFloat16) {
    launch_kd_ce_fwd_bf16(sc.data_ptr(), tc.data_ptr(), lpr.data_ptr<float>(), B, C,
                          cur_stream());
  } else {
    TORCH_CHECK(false, "kd_ce: fp32/bf16 only");
  }
  return lpr;
}

torch::Tensor kd_ce_backward(torch::Tensor s, torch::Tensor t, double gout_over_B) {
  auto sc = s.contiguous();
  auto tc = t.contiguous();
  const int B = (int)s.size(0), C = (int)s.size(1);
  auto ds = torch::empty_like(sc);
  if (s.scalar_type() == torch::kFloat32) {
    launch_kd_ce_bwd_f32(sc.data_ptr<float>(), tc.data_ptr<float>(), ds.data_ptr<float>(),
                         (float)gout_over_B, B, C, cur_stream());
  } else if (s.scalar_type() == torch::kBFloat16) {
    launch_kd_ce_bwd_bf16(sc.data_ptr(), tc.data_ptr(), ds.data_ptr(),
                          (float)gout_over_B, B, C, cur_stream());
  } else {
    TORCH_CHECK(false, "kd_ce: fp32/bf16 only");
  }
  return ds;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_sgd", &fused_sgd,
        "fused flat momentum-SGD update (p,g,m flat fp32; folds grad_scale)");
  m.def("kd_ce_forward", &kd_ce_forward, "KD soft-label CE forward -> per-row loss");
  m.def("kd_ce_backward", &kd_ce_backward, "KD soft-label CE backward -> dlogits");
  m.attr("_arch") = "gfx950";
}
