// gfx950 (CDNA4) batched allocation-sizing kernel.
//
// One 256-thread workgroup (4 wave64) per problem.  The whole analytic
// pipeline of core.create_allocation (service-rate table, log-space
// product-form M/M/1/K probabilities, TTFT/ITL bisection, replica count,
// final per-replica analysis) runs in-workgroup:
//
//   - the cumulative log-service-rate table (K = 11*N doubles) lives in LDS
//     and is built with a chunked parallel scan — every later model
//     evaluation is LDS-bandwidth bound, never HBM;
//   - each model evaluation is ONE strided pass over the significant
//     state window (the normalization max and the ~1e-16 cutoff window
//     both come from closed-form O(log K) searches over the concave
//     log-probability curve) with wave64 __shfl_down reductions;
//     consecutive lanes touch consecutive doubles -> conflict-free
//     ds_read_b64 (bank = (a/4) % 64);
//   - the bisection control flow is uniform across the workgroup (all
//     decisions derive from broadcast reduction results), so the
//     __syncthreads() inside evaluations are safe;
//   - LDS budget: (K + 256 + 32) doubles <= 64 KiB for N <= ~700 (the
//     binding routes larger batch limits to the CPU path).  At N = 256
//     (the collector's default max batch) that is ~25 KiB -> 2+ workgroups
//     per CU, and a fleet-sized batch (hundreds of (server, accelerator)
//     pairs) fills all 256 CUs.
//
// Numerics are double throughout, matching the Python analyzer; math
// parity is covered by tests/test_ops.py (CPU) and the @gpu numerics
// tests.

#include <hip/hip_runtime.h>

#include "queue_core.h"

namespace wva {

// Three workgroup geometries are instantiated:
//   - 256 threads (4 wave64): strided sweeps + LDS cross-wave combine;
//   - 64 threads (1 wave64): barrier-free — the wave executes in lockstep,
//     reductions are pure __shfl_down chains and the combined values are
//     broadcast from lane 0 with __shfl (no LDS round-trip);
//   - 128 threads (2 wave64, "dual"): wave 0 runs the TTFT bisection
//     while wave 1 concurrently runs the ITL bisection, each with its own
//     barrier-free 64-lane evaluations — for the common both-targets
//     case this halves the serial-bisection critical path, which is what
//     bounds small-fleet dispatches (a 192-problem launch cannot fill
//     256 CUs, so per-problem latency IS the dispatch time).
// The launcher auto-selects by state-chain length; the WVA_GPU_THREADS
// env var (64|128|256) overrides, read per launch.

// Reduction scratch layout (doubles, after cum[max_k] in dynamic LDS):
//   red[0..WAVES*5-1]  per-wave partials (S, Ni, Snum, Ninum, eK)

template <int THREADS>
struct WgEval {
  static constexpr int WAVES = THREADS / 64;
  const Parms &p;
  const double *cum;  // LDS, K entries
  double *red;        // LDS scratch
  int K;

  __device__ Stats eval(double lam) const {
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    (void)wave;
    (void)lane;
    const double loglam = log(lam);

    // max of logp via the concavity closed form, and the significant
    // state window via two bisections (queue_core.h): every thread
    // computes the identical O(log K) searches — no sweep, no barrier,
    // and the exp sweep shrinks to the ~window where p(n) > 1e-16
    const int n_star = log_mode_state(p, K, lam);
    double m = log_p(cum, loglam, n_star);
    if (m < 0.0) m = 0.0;
    int n_lo, n_hi;
    state_window(cum, loglam, K, n_star, m, &n_lo, &n_hi);
    if constexpr (WAVES > 1) {
      __syncthreads();  // red may still be read from a previous eval
    }

    // single windowed pass: normalization and moment sums (eK is the
    // state-K boundary term, reduced like the sums — only its owner
    // thread contributes a non-zero partial).  For WAVES == 1 the index
    // is the LANE, not the block thread id: that makes WgEval<64> usable
    // per-wave inside the 128-thread dual kernel (each wave sweeps its
    // own full window in lockstep, reductions stay wave-local).
    const int num = p.max_batch;
    const int idx = (WAVES == 1) ? lane : tid;
    double S = 0.0, Ni = 0.0, Snum = 0.0, Ninum = 0.0, eK = 0.0;
    for (int n = n_lo + idx; n <= n_hi; n += THREADS) {
      double e = exp(log_p(cum, loglam, n) - m);
      S += e;
      Ni += (double)n * e;
      if (n <= num) {
        Snum += e;
        Ninum += (double)n * e;
      }
      if (n == K) eK = e;
    }
    for (int off = 32; off > 0; off >>= 1) {
      S += __shfl_down(S, off, 64);
      Ni += __shfl_down(Ni, off, 64);
      Snum += __shfl_down(Snum, off, 64);
      Ninum += __shfl_down(Ninum, off, 64);
      eK += __shfl_down(eK, off, 64);
    }
    if constexpr (WAVES == 1) {
      // single wave: broadcast lane 0's totals to every lane
      S = __shfl(S, 0, 64);
      Ni = __shfl(Ni, 0, 64);
      Snum = __shfl(Snum, 0, 64);
      Ninum = __shfl(Ninum, 0, 64);
      eK = __shfl(eK, 0, 64);
    } else {
      if (lane == 0) {
        red[wave * 5 + 0] = S;
        red[wave * 5 + 1] = Ni;
        red[wave * 5 + 2] = Snum;
        red[wave * 5 + 3] = Ninum;
        red[wave * 5 + 4] = eK;
      }
      __syncthreads();
      S = Ni = Snum = Ninum = eK = 0.0;
      for (int w = 0; w < WAVES; ++w) {
        S += red[w * 5 + 0];
        Ni += red[w * 5 + 1];
        Snum += red[w * 5 + 2];
        Ninum += red[w * 5 + 3];
        eK += red[w * 5 + 4];
      }
    }

    Stats st;
    st.throughput = lam * (1.0 - eK / S);
    double n_sys = Ni / S;
    st.n_serv = Ninum / S + (1.0 - Snum / S) * (double)num;
    if (st.throughput == 0.0) {
      st.wait = st.serv = 0.0;
    } else {
      double resp = n_sys / st.throughput;
      st.serv = st.n_serv / st.throughput;
      st.wait = resp - st.serv;
      if (st.wait < 0.0) st.wait = 0.0;
    }
    return st;
  }

  __device__ double eval_ttft(double lam) const { return eval_ttft_of(p, eval(lam)); }
  __device__ double eval_itl(double lam) const { return eval_itl_of(p, eval(lam)); }
};

// Uniform-control-flow bisection (all threads run it in lockstep; the eval
// results are identical on every thread because they come from broadcast
// reductions).  Semantics mirror analyzer/search.py.
template <typename F>
__device__ int wg_binary_search(double x_min, double x_max, double y_target, F eval,
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

__device__ inline Parms load_parms(const double *pr) {
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

// chunked parallel inclusive scan of log_mu over K states into LDS
template <int THREADS>
__device__ void build_cum(const Parms &p, int K, double *cum, double *totals) {
  const int tid = threadIdx.x;
  const int chunk = (K + THREADS - 1) / THREADS;
  const int lo = tid * chunk;
  const int hi = min(lo + chunk, K);
  double acc = 0.0;
  for (int n = lo; n < hi; ++n) {
    acc += log_mu(p, n);
    cum[n] = acc;
  }
  totals[tid] = (lo < K) ? acc : 0.0;
  __syncthreads();
  double offset = 0.0;
  for (int t = 0; t < tid; ++t) offset += totals[t];  // broadcast LDS reads
  for (int n = lo; n < hi; ++n) cum[n] += offset;
  __syncthreads();
}

template <int THREADS>
__device__ void solve_body(const double *__restrict__ prob, double *__restrict__ out,
                           int n_problems, int max_k) {
  const int pid = blockIdx.x;
  if (pid >= n_problems) return;
  const double *pr = prob + (size_t)pid * PROBLEM_FIELDS;
  double *res = out + (size_t)pid * RESULT_FIELDS;
  const int tid = threadIdx.x;

  const Parms p = load_parms(pr);
  const int K = p.max_batch * (1 + kMaxQueueToBatchRatio);

  extern __shared__ double smem[];
  double *cum = smem;             // this problem's K entries
  double *totals = smem + max_k;  // THREADS chunk totals
  double *red = totals + THREADS;

  build_cum<THREADS>(p, K, cum, totals);

  const double lam_min = serv_rate(p, 1) * kEpsilon;  // req/ms
  const double lam_max = serv_rate(p, p.max_batch) * (1.0 - kEpsilon);

  WgEval<THREADS> ev{p, cum, red, K};

  if (tid < RESULT_FIELDS && pid < n_problems) res[tid] = 0.0;

  double lam_ttft = lam_max;
  if (pr[P_TARGET_TTFT] > 0.0) {
    int ind = wg_binary_search(
        lam_min, lam_max, pr[P_TARGET_TTFT],
        [&](double x) { return ev.eval_ttft(x); }, &lam_ttft);
    if (ind < 0) return;
  }
  double lam_itl = lam_max;
  if (pr[P_TARGET_ITL] > 0.0) {
    int ind = wg_binary_search(
        lam_min, lam_max, pr[P_TARGET_ITL],
        [&](double x) { return ev.eval_itl(x); }, &lam_itl);
    if (ind < 0) return;
  }
  double lam_tps = lam_max;
  if (pr[P_TARGET_TPS] > 0.0) lam_tps = lam_max * (1.0 - kStabilityFraction);

  const double lam = fmin(lam_ttft, fmin(lam_itl, lam_tps));
  Stats st = ev.eval(lam);
  const double rate_star = st.throughput * 1000.0;  // req/s

  const double total_rate = pr[P_TOTAL_RATE];
  double n_rep = ceil(total_rate / rate_star);
  if (n_rep < pr[P_MIN_REPLICAS]) n_rep = pr[P_MIN_REPLICAS];
  const double rate = total_rate / n_rep;
  if (rate <= 0.0 || rate > lam_max * 1000.0) return;

  Stats fin = ev.eval(rate / 1000.0);
  double rho = fin.n_serv / (double)p.max_batch;
  rho = fmin(fmax(rho, 0.0), 1.0);

  if (tid == 0) {
    res[R_FEASIBLE] = 1.0;
    res[R_REPLICAS] = n_rep;
    res[R_RATE_STAR] = rate_star;
    res[R_ITL] = eval_itl_of(p, fin);
    res[R_TTFT] = eval_ttft_of(p, fin);
    res[R_RHO] = rho;
  }
}

// 2-wave "dual" body: the TTFT and ITL bisections are independent, so
// wave 0 searches the TTFT target while wave 1 searches the ITL target
// concurrently.  Each wave's evaluations are the barrier-free WgEval<64>
// (lane-indexed, __shfl-only), so the waves may diverge arbitrarily
// between the scan barrier and the exchange barrier.  Semantics are
// identical to the sequential body: the original runs TTFT first and
// returns infeasible without touching ITL — here both run, but the
// combined feasibility check and the zeroed result row give the same
// observable output.
__device__ void solve_body_dual(const double *__restrict__ prob, double *__restrict__ out,
                                int n_problems, int max_k) {
  constexpr int THREADS = 128;
  const int pid = blockIdx.x;
  if (pid >= n_problems) return;
  const double *pr = prob + (size_t)pid * PROBLEM_FIELDS;
  double *res = out + (size_t)pid * RESULT_FIELDS;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;

  const Parms p = load_parms(pr);
  const int K = p.max_batch * (1 + kMaxQueueToBatchRatio);

  extern __shared__ double smem[];
  double *cum = smem;             // this problem's K entries
  double *totals = smem + max_k;  // THREADS chunk totals
  double *ex = totals + THREADS;  // exchange: lam_ttft, ind_ttft, lam_itl, ind_itl

  build_cum<THREADS>(p, K, cum, totals);

  const double lam_min = serv_rate(p, 1) * kEpsilon;  // req/ms
  const double lam_max = serv_rate(p, p.max_batch) * (1.0 - kEpsilon);

  WgEval<64> ev{p, cum, nullptr, K};  // lane-indexed, barrier-free

  if (tid < RESULT_FIELDS) res[tid] = 0.0;

  double lam_t = lam_max;
  int ind = 0;
  if (wave == 0) {
    if (pr[P_TARGET_TTFT] > 0.0) {
      ind = wg_binary_search(
          lam_min, lam_max, pr[P_TARGET_TTFT],
          [&](double x) { return ev.eval_ttft(x); }, &lam_t);
    }
  } else {
    if (pr[P_TARGET_ITL] > 0.0) {
      ind = wg_binary_search(
          lam_min, lam_max, pr[P_TARGET_ITL],
          [&](double x) { return ev.eval_itl(x); }, &lam_t);
    }
  }
  if ((tid & 63) == 0) {
    ex[wave * 2 + 0] = lam_t;
    ex[wave * 2 + 1] = (double)ind;
  }
  __syncthreads();
  if (ex[1] < 0.0 || ex[3] < 0.0) return;  // a target below the reachable range
  if (wave == 1) return;                   // wave 0 finishes the tail alone

  double lam_tps = lam_max;
  if (pr[P_TARGET_TPS] > 0.0) lam_tps = lam_max * (1.0 - kStabilityFraction);

  const double lam = fmin(ex[0], fmin(ex[2], lam_tps));
  Stats st = ev.eval(lam);
  const double rate_star = st.throughput * 1000.0;  // req/s

  const double total_rate = pr[P_TOTAL_RATE];
  double n_rep = ceil(total_rate / rate_star);
  if (n_rep < pr[P_MIN_REPLICAS]) n_rep = pr[P_MIN_REPLICAS];
  const double rate = total_rate / n_rep;
  if (rate <= 0.0 || rate > lam_max * 1000.0) return;

  Stats fin = ev.eval(rate / 1000.0);
  double rho = fin.n_serv / (double)p.max_batch;
  rho = fmin(fmax(rho, 0.0), 1.0);

  if (tid == 0) {
    res[R_FEASIBLE] = 1.0;
    res[R_REPLICAS] = n_rep;
    res[R_RATE_STAR] = rate_star;
    res[R_ITL] = eval_itl_of(p, fin);
    res[R_TTFT] = eval_ttft_of(p, fin);
    res[R_RHO] = rho;
  }
}

extern "C" __global__ void __launch_bounds__(256) wva_solve_kernel_256(
    const double *__restrict__ prob, double *__restrict__ out, int n_problems,
    int max_k) {
  solve_body<256>(prob, out, n_problems, max_k);
}

extern "C" __global__ void __launch_bounds__(64) wva_solve_kernel_64(
    const double *__restrict__ prob, double *__restrict__ out, int n_problems,
    int max_k) {
  solve_body<64>(prob, out, n_problems, max_k);
}

extern "C" __global__ void __launch_bounds__(128) wva_solve_kernel_128(
    const double *__restrict__ prob, double *__restrict__ out, int n_problems,
    int max_k) {
  solve_body_dual(prob, out, n_problems, max_k);
}

}  // namespace wva

#include <cstdlib>
#include <cstring>

extern "C" void wva_launch_solve(const double *prob, double *out, int n_problems,
                                 int max_k, void *stream) {
  // Geometry auto-selects on state-chain length (measured:
  // profiles/r01_queue_solver.md, profiles/r01_kernel_sweep.json).  The
  // 2-wave dual kernel overlaps the TTFT and ITL bisections and is the
  // default below the large-K regime; the 4-wave kernel's sweep
  // parallelism takes over for long chains.  WVA_GPU_THREADS=64|128|256
  // overrides (64 = the sequential single-wave kernel).
  const char *env = std::getenv("WVA_GPU_THREADS");
  int threads;
  if (env != nullptr && std::strcmp(env, "256") == 0) {
    threads = 256;
  } else if (env != nullptr && std::strcmp(env, "128") == 0) {
    threads = 128;
  } else if (env != nullptr && std::strcmp(env, "64") == 0) {
    threads = 64;
  } else if (n_problems >= 512) {
    // chip-filling launches: the dual kernel won or tied every measured
    // (B, K) point at B=4096 (incl. K=7700, where it matches 4-wave)
    threads = 128;
  } else {
    // underfilled launches are per-problem latency-bound: overlapping
    // the two bisections wins while evaluations are short, but from
    // K~2816 the 4-wave sweep parallelism matters more (B=192 data)
    threads = max_k < 2048 ? 128 : 256;
  }
  const size_t smem = (size_t)(max_k + threads + 32) * sizeof(double);
  if (threads == 64) {
    hipLaunchKernelGGL(wva::wva_solve_kernel_64, dim3(n_problems), dim3(64), smem,
                       (hipStream_t)stream, prob, out, n_problems, max_k);
  } else if (threads == 128) {
    hipLaunchKernelGGL(wva::wva_solve_kernel_128, dim3(n_problems), dim3(128), smem,
                       (hipStream_t)stream, prob, out, n_problems, max_k);
  } else {
    hipLaunchKernelGGL(wva::wva_solve_kernel_256, dim3(n_problems), dim3(256), smem,
                       (hipStream_t)stream, prob, out, n_problems, max_k);
  }
}
