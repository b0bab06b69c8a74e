// Host-side scalar solve of one sizing problem, shared by the torch
// binding (queue_solver.cpp: CPU at::parallel_for + HIP dispatch glue)
// and the torch-free pybind11/numpy binding (queue_solver_cpu.cpp) used
// in the slim controller image.  Semantics mirror
// core.create_allocation's non-zero-load body (see queue_solver.cpp's
// header comment for the reference citation).
#pragma once

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
inline void solve_one(const double *pr, double *out, double *cum) {
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
