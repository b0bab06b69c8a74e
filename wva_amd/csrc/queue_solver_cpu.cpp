// Torch-free batched allocation sizing: pybind11 + numpy + OpenMP.
//
// The controller container image is slim (no libtorch, no ROCm); this
// binding gives it the same native CPU sizing path as the full build by
// reusing queue_host.h verbatim.  Input/output are numpy float64 arrays
// with the layouts of queue_core.h ([B,12] problems -> [B,6] results).

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <stdexcept>
#include <vector>

#include "queue_host.h"

namespace py = pybind11;

static py::array_t<double> solve_allocations(
    py::array_t<double, py::array::c_style | py::array::forcecast> problems) {
  auto buf = problems.request();
  if (buf.ndim != 2 || buf.shape[1] != (ssize_t)wva::PROBLEM_FIELDS)
    throw std::invalid_argument("problems must be [B, 12] float64");
  const ssize_t B = buf.shape[0];
  auto out = py::array_t<double>({B, (ssize_t)wva::RESULT_FIELDS});
  auto out_buf = out.request();
  const double *pr = (const double *)buf.ptr;
  double *res = (double *)out_buf.ptr;
  for (ssize_t i = 0; i < B * (ssize_t)wva::RESULT_FIELDS; ++i) res[i] = 0.0;
  if (B == 0) return out;

  int max_k = 1;
  for (ssize_t i = 0; i < B; ++i) {
    int k = (int)pr[i * wva::PROBLEM_FIELDS + wva::P_MAX_BATCH] *
            (1 + wva::kMaxQueueToBatchRatio);
    if (k > max_k) max_k = k;
  }

  {
    py::gil_scoped_release release;
#pragma omp parallel
    {
      std::vector<double> cum((size_t)max_k);
#pragma omp for schedule(dynamic, 8)
      for (ssize_t i = 0; i < B; ++i) {
        wva::solve_one(pr + i * wva::PROBLEM_FIELDS, res + i * wva::RESULT_FIELDS,
                       cum.data());
      }
    }
  }
  return out;
}

PYBIND11_MODULE(_queue_native_cpu, m) {
  m.doc() = "wva_amd torch-free batched queue solver (CPU/OpenMP)";
  m.def("solve_allocations", &solve_allocations,
        "Batched state-dependent M/M/1/K allocation sizing", py::arg("problems"));
  m.attr("PROBLEM_FIELDS") = (int)wva::PROBLEM_FIELDS;
  m.attr("RESULT_FIELDS") = (int)wva::RESULT_FIELDS;
}
