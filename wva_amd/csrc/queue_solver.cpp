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

#include "queue_core.h"

namespace wva {

// Monotone bisection mirroring analyzer/search.py (boundary classification
// -1 below / 0 within / +1 above, relative tolerance, 100 iterations).
template <typename F>
inline int binary_search_host(double x_min, double x_max, double y_target, F eval,
                              double *x_star) {
  double y0 = eval(x_min);
  if (within_tolerance(y0, y_target, kTolerance)) {
    *x_star = x_min;
    return 0;
  }
  double y1 = eval(x_max);
  if (within_tolerance(y1, y_target, kTolerance)) {
    *x_star = x_max;
    return 0;
  }
  if (within_tolerance(y0, y1, kTolerance)) {
    // flat function: classify by value only (direction would be noise)
    if (y_target > fmax(y0, y1)) {
      *x_star = x_max;
      return +1;
    }
    *x_star = x_min;
    return -1;
  }
  bool increasing = y0 < y1;
  if ((increasing && y_target < y0) || (!increasing && y_target > y0)) {
    *x_star = x_min;
    return -1;
  }
  if ((increasing && y_target > y1) || (!increasing && y_target < y1)) {
    *x_star = x_max;
    return +1;
  }
  double xs = x_min;
  for (int i = 0; i < kMaxIterations; ++i) {
    xs = 0.5 * (x_min + x_max);
    double ys = eval(xs);
    if (within_tolerance(ys, y_target, kTolerance)) break;
    if ((increasing && y_target < ys) || (!increasing && y_target > ys)) {
      x_max = xs;
    } else {
      x_min = xs;
    }
  }
  *x_star = xs;
  return 0;
}

inline Parms parms_from_problem(const double *pr) {
  Parms p;
  p.alpha = pr[P_ALPHA];
  p.beta = pr[P_BETA];
  p.gamma = pr[P_GAMMA];
  p.delta = pr[P_DELTA];
  p.in_tokens = pr[P_IN_TOKENS];
  p.out_tokens = (int)pr[P_OUT_TOKENS];
  p.max_batch = (int)pr[P_MAX_BATCH];
  p.num_decode = p.out_tokens - 1;
  if (p.in_tokens == 0.0 && p.out_tokens == 1) p.num_decode = 1;
  return p;
}

// Scalar solve of one problem (CPU path).  cum must hold K doubles.
static void solve_one(const double *pr, double *out, double *cum) {
  Parms p = parms_from_problem(pr);
  const int K = p.max_batch * (1 + kMaxQueueToBatchRatio);

  double acc = 0.0;
  for (int n = 0; n < K; ++n) {
    acc += log_mu(p, n);
    cum[n] = acc;
  }
  const double lam_min = serv_rate(p, 1) * kEpsilon;           // req/ms
  const double lam_max = serv_rate(p, p.max_batch) * (1.0 - kEpsilon);

  auto eval_ttft = [&](double lam) { return eval_ttft_of(p, eval_model(p, cum, K, lam)); };
  auto eval_itl = [&](double lam) { return eval_itl_of(p, eval_model(p, cum, K, lam)); };

  for (int f = 0; f < RESULT_FIELDS; ++f) out[f] = 0.0;

  double lam_ttft = lam_max;
  if (pr[P_TARGET_TTFT] > 0.0) {
    if (binary_search_host(lam_min, lam_max, pr[P_TARGET_TTFT], eval_ttft, &lam_ttft) < 0)
      return;  // target below the bounded region: infeasible
  }
  double lam_itl = lam_max;
  if (pr[P_TARGET_ITL] > 0.0) {
    if (binary_search_host(lam_min, lam_max, pr[P_TARGET_ITL], eval_itl, &lam_itl) < 0)
      return;
  }
  double lam_tps = lam_max;
  if (pr[P_TARGET_TPS] > 0.0) lam_tps = lam_max * (1.0 - kStabilityFraction);

  double lam = fmin(lam_ttft, fmin(lam_itl, lam_tps));
  Stats st = eval_model(p, cum, K, lam);
  const double rate_star = st.throughput * 1000.0;  // req/s

  const double total_rate = pr[P_TOTAL_RATE];
  double n_rep = ceil(total_rate / rate_star);
  if (n_rep < pr[P_MIN_REPLICAS]) n_rep = pr[P_MIN_REPLICAS];
  const double rate = total_rate / n_rep;  // req/s per replica
  if (rate <= 0.0 || rate > lam_max * 1000.0) return;

  Stats fin = eval_model(p, cum, K, rate / 1000.0);
  double rho = fin.n_serv / (double)p.max_batch;
  if (rho < 0.0) rho = 0.0;
  if (rho > 1.0) rho = 1.0;

  out[R_FEASIBLE] = 1.0;
  out[R_REPLICAS] = n_rep;
  out[R_RATE_STAR] = rate_star;
  out[R_ITL] = eval_itl_of(p, fin);
  out[R_TTFT] = eval_ttft_of(p, fin);
  out[R_RHO] = rho;
}

}  // namespace wva

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
