// Python bindings for the gfx950 gossip kernels (HIP-native; uses the
// c10::hip stream API directly — no CUDA-compat layer).

#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

extern "C" {
void sgp_scale(float* x, const float* a, int64_t n, hipStream_t stream);
void sgp_add_scale(float* x, const float* r, const float* a, int64_t n,
                   hipStream_t stream);
void sgp_pack_mix(float* x, float* out, const float* a, int64_t n,
                  hipStream_t stream);
void sgp_average(float* x, const float* y, int64_t n, hipStream_t stream);
void sgp_sgd_step(float* p, const float* g, float* buf, double lr, double mu,
                  double wd, double damp, bool nesterov, bool first, int64_t n,
                  hipStream_t stream);
}

namespace {

hipStream_t current_stream(const torch::Tensor& t) {
  return c10::hip::getCurrentHIPStream(t.device().index()).stream();
}

void check_flat(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a device tensor");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

void check_scalar(const torch::Tensor& a, const torch::Tensor& like) {
  TORCH_CHECK(a.is_cuda(), "scalar must be a device tensor");
  TORCH_CHECK(a.numel() == 1, "scalar must have exactly one element");
  TORCH_CHECK(a.scalar_type() == torch::kFloat32, "scalar must be fp32");
  TORCH_CHECK(a.device() == like.device(), "scalar on wrong device");
}

void scale_(torch::Tensor x, torch::Tensor a) {
  check_flat(x, "x");
  check_scalar(a, x);
  sgp_scale(x.data_ptr<float>(), a.data_ptr<float>(), x.numel(),
            current_stream(x));
}

void add_scale_(torch::Tensor x, torch::Tensor r, torch::Tensor a) {
  check_flat(x, "x");
  check_flat(r, "r");
  check_scalar(a, x);
  TORCH_CHECK(x.numel() == r.numel(), "size mismatch");
  sgp_add_scale(x.data_ptr<float>(), r.data_ptr<float>(), a.data_ptr<float>(),
                x.numel(), current_stream(x));
}

void pack_mix_(torch::Tensor x, torch::Tensor out, torch::Tensor a) {
  check_flat(x, "x");
  check_flat(out, "out");
  check_scalar(a, x);
  TORCH_CHECK(x.numel() == out.numel(), "size mismatch");
  sgp_pack_mix(x.data_ptr<float>(), out.data_ptr<float>(), a.data_ptr<float>(),
               x.numel(), current_stream(x));
}

void average_(torch::Tensor x, torch::Tensor y) {
  check_flat(x, "x");
  check_flat(y, "y");
  TORCH_CHECK(x.numel() == y.numel(), "size mismatch");
  sgp_average(x.data_ptr<float>(), y.data_ptr<float>(), x.numel(),
              current_stream(x));
}

void sgd_step_(torch::Tensor p, torch::Tensor g, torch::Tensor buf, double lr,
               double momentum, double weight_decay, double dampening,
               bool nesterov, bool first_step) {
  check_flat(p, "p");
  check_flat(g, "g");
  check_flat(buf, "buf");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == buf.numel(),
              "size mismatch");
  sgp_sgd_step(p.data_ptr<float>(), g.data_ptr<float>(), buf.data_ptr<float>(),
               lr, momentum, weight_decay, dampening, nesterov, first_step,
               p.numel(), current_stream(p));
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "gfx950 fused gossip kernels";
  m.def("scale_", &scale_, "x *= a (in place, fused over flat buffer)");
  m.def("add_scale_", &add_scale_, "x = (x + r) * a");
  m.def("pack_mix_", &pack_mix_, "x *= a; out = x");
  m.def("average_", &average_, "x = (x + y) / 2");
  m.def("sgd_step_", &sgd_step_, "fused momentum-SGD step");
}
