// Batched allocation sizing: torch binding + CPU (at::parallel_for) path.
//
// One "problem" is the non-zero-load body of core.create_allocation
// (/root/reference/pkg/core/allocation.go:77-160 re-designed): given the
// linear perf parameters, request shape, batch limit and SLO targets,
// find the max per-replica rate meeting the targets, the replica count for
// the offered load, and the predicted ITL/TTFT/rho at the final rate.
//
// The GPU path (queue_solver_hip.hip) runs one 256-thread workgroup per
// problem on gfx950; this file provides the same math on CPU and the
// dispatch glue.

#include <torch/extension.h>

#ifdef WVA_WITH_HIP
#include <c10/hip/HIPStream.h>
#endif

#include <cmath>
#include <vector>


// Host scalar solve + bisection shared with the torch-free binding.
#include "queue_host.h"

#ifdef WVA_WITH_HIP
extern "C" void wva_launch_solve(const double *prob, double *out, int n_problems,
                                 int max_k, void *stream);
#endif

static torch::Tensor solve_allocations(torch::Tensor problems, int64_t max_k_hint = -1) {
  TORCH_CHECK(problems.dim() == 2 && problems.size(1) == wva::PROBLEM_FIELDS,
              "problems must be [B, ", (int)wva::PROBLEM_FIELDS, "]");
  TORCH_CHECK(problems.scalar_type() == torch::kFloat64, "problems must be float64");
  problems = problems.contiguous();
  const int64_t B = problems.size(0);
  auto out = torch::zeros({B, (int64_t)wva::RESULT_FIELDS}, problems.options());
  if (B == 0) return out;

  if (problems.is_cuda()) {
#ifdef WVA_WITH_HIP
    int max_k;
    if (max_k_hint > 0) {
      // caller-supplied chain length (hipGraph capture cannot tolerate
      // the .item() device sync below; wva_amd/ops/batched.py computes
      // it host-side from the numpy batch before upload)
      max_k = (int)max_k_hint;
    } else {
      const double max_batch = problems.select(1, wva::P_MAX_BATCH).max().item<double>();
      max_k = (int)max_batch * (1 + wva::kMaxQueueToBatchRatio);
    }
    TORCH_CHECK((max_k + 288) * 8 <= 64 * 1024,
                "max_batch too large for the LDS-resident GPU path (limit ~700); "
                "use the CPU path for these problems");
    auto stream = c10::hip::getCurrentHIPStream();
    wva_launch_solve(problems.data_ptr<double>(), out.data_ptr<double>(), (int)B,
                     max_k, (void *)stream.stream());
    return out;
#else
    TORCH_CHECK(false, "wva native extension built without HIP support");
#endif
  }

  const double *pr = problems.data_ptr<double>();
  double *res = out.data_ptr<double>();
  // scratch sized by the largest K in the batch
  int max_k = 1;
  for (int64_t i = 0; i < B; ++i) {
    int k = (int)pr[i * wva::PROBLEM_FIELDS + wva::P_MAX_BATCH] *
            (1 + wva::kMaxQueueToBatchRatio);
    if (k > max_k) max_k = k;
  }
  at::parallel_for(0, B, 1, [&](int64_t begin, int64_t end) {
    std::vector<double> cum((size_t)max_k);
    for (int64_t i = begin; i < end; ++i) {
      wva::solve_one(pr + i * wva::PROBLEM_FIELDS, res + i * wva::RESULT_FIELDS,
                     cum.data());
    }
  });
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "wva_amd native batched queue solver (CPU + gfx950 HIP)";
  m.def("solve_allocations", &solve_allocations,
        "Batched state-dependent M/M/1/K allocation sizing",
        pybind11::arg("problems"), pybind11::arg("max_k_hint") = -1);
  m.attr("PROBLEM_FIELDS") = (int)wva::PROBLEM_FIELDS;
  m.attr("RESULT_FIELDS") = (int)wva::RESULT_FIELDS;
#ifdef WVA_WITH_HIP
  m.attr("HAS_HIP") = true;
#else
  m.attr("HAS_HIP") = false;
#endif
}
