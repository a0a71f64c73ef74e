// PyTorch bindings for the sagecal_amd gfx950 kernel library.
// Host-compiled: all kernel launches live in launchers.hip (hipcc).
#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

extern "C" {
hipError_t launch_predict_coh(const double*, const double*, const double*,
    const double*, const double*, const double*, const float*,
    const float*, const float*, const float*, const float*, const float*,
    const float*, const float*, const float*, const float*, const float*,
    const float*, const int*, const int*, int, int, double, double,
    double, const float*, const int*, int, int, int, float2*,
    hipStream_t);
hipError_t launch_jtj_accum(const float2*, const float2*, const float2*,
    const int*, const int*, const float*, int, int, int, int, float2*,
    float2*, float2*, float*, int, int, hipStream_t);
hipError_t launch_jtj_expand(const float2*, const float2*, const float2*,
    const int*, int, int, int, float*, float*, hipStream_t);
hipError_t launch_model_cost(const float2*, const float2*, const float2*,
    const int*, const int*, const float*, int, int, int, int, float*,
    hipStream_t);
hipError_t launch_apply_jones(const float2*, const float2*, const float2*,
    const int*, const int*, int, int, int, int, int, int, float2*,
    hipStream_t);
hipError_t launch_chol_mw(const float*, const float*, const float*, int,
                          int, float*, float*, int*, int, hipStream_t);
hipError_t launch_chol_solve(const float*, const float*, const float*, int,
    int, float*, float*, int*, int, hipStream_t);
}

#define CHECK_HIP(x) do { hipError_t e = (x); TORCH_CHECK(e == hipSuccess, \
    "HIP error: ", hipGetErrorString(e)); } while (0)

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

static float2* cptr(torch::Tensor& t) {
  return reinterpret_cast<float2*>(t.data_ptr());
}
static const float2* ccptr(const torch::Tensor& t) {
  return reinterpret_cast<const float2*>(t.data_ptr());
}

torch::Tensor predict_coh(
    torch::Tensor u, torch::Tensor v, torch::Tensor w,
    torch::Tensor ll, torch::Tensor mm, torch::Tensor nn1,
    torch::Tensor sI, torch::Tensor sQ, torch::Tensor sU, torch::Tensor sV,
    torch::Tensor eX, torch::Tensor eY, torch::Tensor eP,
    torch::Tensor cxi, torch::Tensor sxi, torch::Tensor cphi,
    torch::Tensor sphi, torch::Tensor r1, torch::Tensor stype,
    torch::Tensor cluster_off, double freq, double fdelta2, double tdelta,
    c10::optional<torch::Tensor> beam, c10::optional<torch::Tensor> pairs,
    int64_t Nbase) {
  const int R = u.size(0);
  const int M = cluster_off.size(0) - 1;
  auto out = torch::empty({M, R, 4},
      torch::dtype(torch::kComplexFloat).device(u.device()));
  const float* beam_p = nullptr;
  const int* pairs_p = nullptr;
  int Ktot = 0, Nsta = 0;
  if (beam.has_value()) {
    TORCH_CHECK(pairs.has_value(), "beam needs the station-pair table");
    TORCH_CHECK(beam->is_contiguous() && beam->dim() == 3,
                "beam must be contiguous [T, K, N] float32");
    beam_p = beam->data_ptr<float>();
    pairs_p = pairs->data_ptr<int>();
    Ktot = beam->size(1);
    Nsta = beam->size(2);
  }
  CHECK_HIP(launch_predict_coh(
      u.data_ptr<double>(), v.data_ptr<double>(), w.data_ptr<double>(),
      ll.data_ptr<double>(), mm.data_ptr<double>(), nn1.data_ptr<double>(),
      sI.data_ptr<float>(), sQ.data_ptr<float>(), sU.data_ptr<float>(),
      sV.data_ptr<float>(), eX.data_ptr<float>(), eY.data_ptr<float>(),
      eP.data_ptr<float>(), cxi.data_ptr<float>(), sxi.data_ptr<float>(),
      cphi.data_ptr<float>(), sphi.data_ptr<float>(), r1.data_ptr<float>(),
      stype.data_ptr<int>(), cluster_off.data_ptr<int>(), M, R, freq,
      fdelta2, tdelta, beam_p, pairs_p, (int)Nbase, Ktot, Nsta,
      cptr(out), cur_stream()));
  return out;
}

std::vector<torch::Tensor> jtj_jtr(
    torch::Tensor x, torch::Tensor coh, torch::Tensor J,
    torch::Tensor pairs, torch::Tensor chunk_tab, torch::Tensor pidx,
    c10::optional<torch::Tensor> wts, int64_t Nbase, int64_t T, int64_t N,
    int64_t nseg, int64_t Mt) {
  auto fopts = torch::dtype(torch::kFloat).device(x.device());
  auto copts = torch::dtype(torch::kComplexFloat).device(x.device());
  const int64_t npair = Nbase;
  auto D = torch::zeros({Mt * N, 4}, copts);
  auto g = torch::zeros({Mt * N, 4}, copts);
  auto Cx = torch::zeros({Mt * npair, 16}, copts);
  auto cost = torch::zeros({Mt}, fopts);
  const float* wp = wts.has_value() ? wts->data_ptr<float>() : nullptr;
  CHECK_HIP(launch_jtj_accum(ccptr(x), ccptr(coh), ccptr(J),
      pairs.data_ptr<int>(), chunk_tab.data_ptr<int>(), wp,
      (int)Nbase, (int)T, (int)N, (int)nseg, cptr(D), cptr(g), cptr(Cx),
      cost.data_ptr<float>(), (int)npair, 0, cur_stream()));
  auto JtJ = torch::empty({Mt, 8 * N, 8 * N}, fopts);
  auto Jtr = torch::empty({Mt, 8 * N}, fopts);
  CHECK_HIP(launch_jtj_expand(ccptr(D), ccptr(g), ccptr(Cx),
      pidx.data_ptr<int>(), (int)N, (int)npair, (int)Mt,
      JtJ.data_ptr<float>(), Jtr.data_ptr<float>(), cur_stream()));
  return {JtJ, Jtr, cost};
}

torch::Tensor model_cost(
    torch::Tensor x, torch::Tensor coh, torch::Tensor J,
    torch::Tensor pairs, torch::Tensor chunk_tab,
    c10::optional<torch::Tensor> wts, int64_t Nbase, int64_t T, int64_t N,
    int64_t nseg, int64_t Mt) {
  auto cost = torch::zeros({Mt},
      torch::dtype(torch::kFloat).device(x.device()));
  const float* wp = wts.has_value() ? wts->data_ptr<float>() : nullptr;
  CHECK_HIP(launch_model_cost(ccptr(x), ccptr(coh), ccptr(J),
      pairs.data_ptr<int>(), chunk_tab.data_ptr<int>(), wp,
      (int)Nbase, (int)T, (int)N, (int)nseg, cost.data_ptr<float>(),
      cur_stream()));
  return cost;
}

torch::Tensor apply_jones(
    c10::optional<torch::Tensor> x, torch::Tensor cohs, torch::Tensor J,
    torch::Tensor pairs, torch::Tensor chunk_tab, int64_t Nbase, int64_t T,
    int64_t N, int64_t nseg, int64_t M, int64_t sub) {
  const int64_t B = nseg * T * Nbase;
  auto out = torch::empty({B, 4},
      torch::dtype(torch::kComplexFloat).device(cohs.device()));
  const float2* xp = x.has_value() ? ccptr(*x) : nullptr;
  CHECK_HIP(launch_apply_jones(xp, ccptr(cohs), ccptr(J),
      pairs.data_ptr<int>(), chunk_tab.data_ptr<int>(), (int)Nbase, (int)T,
      (int)N, (int)nseg, (int)M, (int)sub, cptr(out), cur_stream()));
  return out;
}

std::vector<torch::Tensor> chol_solve(
    torch::Tensor JtJ, torch::Tensor Jtr, torch::Tensor mu,
    torch::Tensor scratch, int64_t stages) {
  const int64_t batch = JtJ.size(0);
  const int64_t n = JtJ.size(1);
  auto dp = torch::empty({batch, n},
      torch::dtype(torch::kFloat).device(JtJ.device()));
  auto info = torch::zeros({batch},
      torch::dtype(torch::kInt).device(JtJ.device()));
  CHECK_HIP(launch_chol_solve(JtJ.data_ptr<float>(), Jtr.data_ptr<float>(),
      mu.data_ptr<float>(), (int)n, (int)batch, scratch.data_ptr<float>(),
      dp.data_ptr<float>(), info.data_ptr<int>(), (int)stages,
      cur_stream()));
  return {dp, info};
}

std::vector<torch::Tensor> chol_solve_mw(
    torch::Tensor JtJ, torch::Tensor Jtr, torch::Tensor mu,
    torch::Tensor scratch, int64_t stages) {
  // multi-workgroup right-looking path: same contract as chol_solve
  const int64_t batch = JtJ.size(0);
  const int64_t n = JtJ.size(1);
  auto dp = torch::empty({batch, n},
      torch::dtype(torch::kFloat).device(JtJ.device()));
  auto info = torch::zeros({batch},
      torch::dtype(torch::kInt).device(JtJ.device()));
  CHECK_HIP(launch_chol_mw(JtJ.data_ptr<float>(), Jtr.data_ptr<float>(),
      mu.data_ptr<float>(), (int)n, (int)batch, scratch.data_ptr<float>(),
      dp.data_ptr<float>(), info.data_ptr<int>(), (int)stages,
      cur_stream()));
  return {dp, info};
}

std::vector<torch::Tensor> jtr_grad(
    torch::Tensor x, torch::Tensor coh, torch::Tensor J,
    torch::Tensor pairs, torch::Tensor chunk_tab,
    c10::optional<torch::Tensor> wts, int64_t Nbase, int64_t T, int64_t N,
    int64_t nseg, int64_t Mt) {
  // gradient-only accumulation (LBFGS path): returns (g [Mt*N,4] c64,
  // cost [Mt]); g's interleaved memory IS vecR order.
  auto fopts = torch::dtype(torch::kFloat).device(x.device());
  auto copts = torch::dtype(torch::kComplexFloat).device(x.device());
  auto g = torch::zeros({Mt * N, 4}, copts);
  auto cost = torch::zeros({Mt}, fopts);
  const float* wp = wts.has_value() ? wts->data_ptr<float>() : nullptr;
  CHECK_HIP(launch_jtj_accum(ccptr(x), ccptr(coh), ccptr(J),
      pairs.data_ptr<int>(), chunk_tab.data_ptr<int>(), wp,
      (int)Nbase, (int)T, (int)N, (int)nseg, nullptr, cptr(g), nullptr,
      cost.data_ptr<float>(), (int)Nbase, 1, cur_stream()));
  return {g, cost};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("jtr_grad", &jtr_grad, "gradient-only accumulation (gfx950)");
  m.def("chol_solve", &chol_solve, "batched damped Cholesky solve (gfx950)");
  m.def("chol_solve_mw", &chol_solve_mw,
        "multi-workgroup damped Cholesky solve (gfx950)");
  m.def("predict_coh", &predict_coh, "coherency predict (gfx950)",
        py::arg("u"), py::arg("v"), py::arg("w"), py::arg("ll"),
        py::arg("mm"), py::arg("nn1"), py::arg("sI"), py::arg("sQ"),
        py::arg("sU"), py::arg("sV"), py::arg("eX"), py::arg("eY"),
        py::arg("eP"), py::arg("cxi"), py::arg("sxi"), py::arg("cphi"),
        py::arg("sphi"), py::arg("r1"), py::arg("stype"),
        py::arg("cluster_off"), py::arg("freq"), py::arg("fdelta2"),
        py::arg("tdelta"), py::arg("beam") = py::none(),
        py::arg("pairs") = py::none(), py::arg("Nbase") = 0);
  m.def("jtj_jtr", &jtj_jtr, "fused JtJ/Jtr assembly (gfx950)");
  m.def("model_cost", &model_cost, "per-chunk model cost (gfx950)");
  m.def("apply_jones", &apply_jones, "model apply / residual (gfx950)");
}
