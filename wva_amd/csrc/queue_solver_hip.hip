// gfx950 (CDNA4) batched allocation-sizing kernel.
//
// One workgroup per problem: the whole analytic pipeline of
// core.create_allocation (service-rate table, log-space product-form
// M/M/1/K probabilities, TTFT/ITL bisection, replica count, final
// per-replica analysis) runs in-workgroup:
//
//   - the cumulative log-service-rate table (K = 11*N doubles) lives in
//     LDS and is built with a chunked parallel scan — every later model
//     evaluation is LDS-bound, never HBM; consecutive lanes touch
//     consecutive doubles -> conflict-free ds_read_b64;
//   - a model evaluation sweeps only the <= N+1 pre-batch states (wave64
//     __shfl_down reductions, 2x-unrolled exp); the whole constant-rate
//     queue region [N+1, K] — 10/11 of the states — sums in CLOSED FORM
//     (geo_tail in queue_core.h, expm1-based geometric sums), which also
//     made the 1e-16 significance window obsolete;
//   - the normalization mode comes from a wave-parallel 64-probe ballot
//     search (wave_lower_bound) instead of a serial bisection;
//   - the default spec geometry (384 = 2 search groups x 3 wave64) runs
//     the TTFT and ITL bisections concurrently AND speculatively: each
//     round a group evaluates its bracket's full depth-2 midpoint tree
//     (bitwise the sequential iterates) and consumes two bisection
//     levels; wave sync is group-local LDS mailboxes with a watchdog —
//     no workgroup barriers after the scan;
//   - LDS budget: (max_k + THREADS + 64) doubles <= 64 KiB for N <= ~700
//     (the binding routes larger batch limits to the CPU path).
//
// Numerics are double throughout, matching the Python analyzer; parity
// is enforced by tests/test_ops.py, the @gpu numerics tests and
// tools/fuzz_parity.py (20k problems, zero feasibility flips).  Design
// narrative + measurements: docs/design/native-queue-solver.md,
// profiles/r02_kernel_notes.md.

#include <hip/hip_runtime.h>

#include "queue_core.h"

namespace wva {

// Five workgroup geometries are instantiated:
//   - 64 (sequential single wave, barrier-free __shfl reductions);
//   - 128 ("dual": wave 0 searches TTFT while wave 1 searches ITL —
//     the winner for chip-filling launches, B >= 512);
//   - 256 (4-wave strided sweeps + LDS cross-wave combine);
//   - 384 (speculative-tree multisection, depth 2 — the default for
//     underfilled, latency-bound launches);
//   - 896 (depth-3 speculation; measured slower than depth 2).
// The launcher auto-selects (occupancy-aware, see wva_launch_solve);
// WVA_GPU_THREADS=64|128|256|384|896 overrides, read per launch.

// Reduction scratch layout (doubles, after cum[max_k] in dynamic LDS):
//   red[0..WAVES*5-1]  per-wave partials (S, Ni, Snum, Ninum, eK)

// Wave-parallel lower bound: smallest i in [lo, hi] with pred(i) true,
// for a monotone (false...false true...true) predicate with pred(hi)
// guaranteed true.  64 probes per round shrink the range 64x, replacing
// ~11 DEPENDENT binary-search iterations (each an LDS-latency chain)
// with ~2 ballot rounds of PARALLEL probes.  Exact: the boundary of a
// monotone predicate is unique, so any probe strategy lands on the same
// index the sequential bisection finds.
template <typename Pred>
__device__ inline int wave_lower_bound(int lo, int hi, int lane, Pred pred) {
  while (hi - lo >= 64) {
    const int step = (hi - lo + 63) >> 6;
    const int i = lo + lane * step;  // lane 0 probes lo itself
    const bool t = (i >= hi) ? true : pred(i);
    const unsigned long long mask = __ballot(t);
    if (mask == 0ull) {
      // every probe (all below hi) is false: boundary past the last one
      lo = lo + 63 * step + 1;
      continue;
    }
    const int first = __ffsll((long long)mask) - 1;
    if (first == 0) return lo;  // pred(lo) already true
    lo = lo + (first - 1) * step + 1;
    const int cap = lo - 1 + step;  // == previous lo + first*step
    if (cap < hi) hi = cap;
  }
  const int i = lo + lane;
  const bool t = (i >= hi) ? true : pred(i);
  const unsigned long long mask = __ballot(t);
  return lo + __ffsll((long long)mask) - 1;
}

template <int THREADS>
struct WgEval {
  static constexpr int WAVES = THREADS / 64;
  const Parms &p;
  const double *cum;  // LDS, K entries
  double *red;        // LDS scratch
  int K;

  // Wave-parallel equivalents of log_mode_state/state_window
  // (queue_core.h): same boundary indices, found with ballots instead of
  // per-lane serial bisections.
  __device__ int mode_state_wave(double lam, int lane) const {
    double tN = prefill_time(p, (double)p.max_batch) +
                p.num_decode * decode_time(p, (double)p.max_batch);
    if ((double)p.max_batch <= lam * tN) return K;
    double t1 = prefill_time(p, 1.0) + p.num_decode * decode_time(p, 1.0);
    if (lam * t1 <= 1.0) return 0;
    const int b = wave_lower_bound(
        1, p.max_batch, lane, [&](int i) { return !rate_below(p, i, lam); });
    return b - 1;
  }

  __device__ Stats eval(double lam) const {
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    (void)wave;
    (void)lane;
    const double loglam = log(lam);

    // max of logp via the concavity closed form — wave-parallel
    // (ballot) in the single-wave geometry, scalar otherwise.  No
    // windowing: the pre-batch region is at most N+1 states (sub-
    // threshold terms just underflow toward 0 in the sweep, adding
    // accuracy, not cost) and the whole queue region [N+1, K] sums in
    // closed form below — so the two window boundary searches the
    // windowed design needed are gone entirely.
    int n_star;
    double m;
    if constexpr (WAVES == 1) {
      n_star = mode_state_wave(lam, lane);
    } else {
      n_star = log_mode_state(p, K, lam);
    }
    m = log_p(cum, loglam, n_star);
    if (m < 0.0) m = 0.0;
    const int n_lo = 0;
    const int n_hi = K;
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
    // Geometric tail over the constant-service-rate queue region
    // (queue_core.h geo_tail): closed-form sum of the states past N,
    // which are up to 10/11 of the window — the per-state exp calls
    // they would cost are the kernel's dominant fp64 work.  Uniform
    // across lanes; added to the reduced totals below.
    const int g_start = (num + 1 > n_lo) ? num + 1 : n_lo;
    GeoTail tail{0.0, 0.0, 0.0, false};
    int loop_hi = n_hi;
    if (g_start + 8 <= n_hi) {
      const double d = loglam - (cum[g_start] - cum[g_start - 1]);
      tail = geo_tail(cum, loglam, d, m, g_start, n_hi, K);
      if (tail.used) loop_hi = g_start - 1;
    }
    double S = 0.0, Ni = 0.0, Snum = 0.0, Ninum = 0.0, eK = 0.0;
    // 2x-unrolled sweep: two independent exp chains in flight per lane
    // hide the fp64 transcendental latency the PMC profile flagged
    // (~34% issue-stall); accumulators are kept separate and combined
    // after the loop so the FLOATING-POINT SUM ORDER PER LANE CHANGES —
    // covered by the cross-geometry ulp-tolerance in the parity suite.
    {
      double S1 = 0.0, Ni1 = 0.0, Snum1 = 0.0, Ninum1 = 0.0, eK1 = 0.0;
      int n = n_lo + idx;
      for (; n + THREADS <= loop_hi; n += 2 * THREADS) {
        const int n2 = n + THREADS;
        const double e = exp(log_p(cum, loglam, n) - m);
        const double e2 = exp(log_p(cum, loglam, n2) - m);
        S += e;
        Ni += (double)n * e;
        S1 += e2;
        Ni1 += (double)n2 * e2;
        if (n <= num) {
          Snum += e;
          Ninum += (double)n * e;
        }
        if (n2 <= num) {
          Snum1 += e2;
          Ninum1 += (double)n2 * e2;
        }
        if (n == K) eK = e;
        if (n2 == K) eK1 = e2;
      }
      if (n <= loop_hi) {
        const double e = exp(log_p(cum, loglam, n) - m);
        S += e;
        Ni += (double)n * e;
        if (n <= num) {
          Snum += e;
          Ninum += (double)n * e;
        }
        if (n == K) eK = e;
      }
      S += S1;
      Ni += Ni1;
      Snum += Snum1;
      Ninum += Ninum1;
      eK += eK1;
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
    // geometric tail joins after the reduction (uniform on all lanes)
    S += tail.S;
    Ni += tail.Ni;
    eK += tail.eK;

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

// ---------------------------------------------------------------------------
// Speculative-tree multisection ("spec") kernel — round-2 lever
// (profiles/r01_rocprof_pmc.txt: ~32% of wave time parked at barriers and
// ~34% issue-stalled on the fp64 exp/log chain of ~200 SEQUENTIAL model
// evaluations; the bisection itself is the critical path).
//
// Geometry: 2 search groups (TTFT, ITL) x SW waves, SW = 2^D - 1.  Each
// round every group evaluates the ENTIRE depth-D midpoint tree of its
// current bracket concurrently — candidate k (heap order) is the
// midpoint reached after the D halvings encoded by k's bits, computed by
// the same (lo+hi)/2 recurrence the sequential loop uses, so the
// iterates are BITWISE identical to wg_binary_search and parity is
// exact, not approximate.  The group then replays the sequential
// decision walk on the cached y values, consuming D bisection levels per
// round: D=2 (SW=3, 384 threads) turns ~100 serial evaluations per
// target into ~50 rounds, with 6 resident waves per CU hiding the exp
// chain latency the PMC profile flagged.
//
// Wave synchronization is group-local LDS mailboxes (publish y, fence,
// bump flag; consumers spin with s_sleep) — NO workgroup barriers after
// the scan, so the two searches never lock-step each other (the dual
// kernel's exchange barrier is gone too).
// ---------------------------------------------------------------------------

__device__ inline void mb_publish(volatile double *slot, double y,
                                  volatile double *flag, double round_id) {
  *slot = y;
  __threadfence_block();
  *flag = round_id;
}

// Watchdog: a legitimate wait is bounded by a sibling wave's evaluation
// (microseconds); ~16M sleep iterations is on the order of a second of
// GPU time.  If a (hypothetical) control-flow divergence bug ever broke
// the uniform-rounds invariant, the spin gives up instead of hanging
// the device — every wave of the group times out the same way, all
// fall through, and the problem's zeroed result row reads as
// infeasible: wrong-but-finite beats a dead GPU.
constexpr unsigned kMailboxSpinLimit = 1u << 24;

__device__ inline bool mb_wait(volatile double *flags, int n, double round_id) {
  for (int j = 0; j < n; ++j) {
    unsigned spins = 0;
    while (flags[j] < round_id) {
      __builtin_amdgcn_s_sleep(2);
      if (++spins > kMailboxSpinLimit) return false;
    }
  }
  __threadfence_block();
  return true;
}

// Midpoint at heap node k (1-based: 1 = root midpoint, 2/3 = children,
// 4..7 = grandchildren).  The bits of k below its leading 1 encode the
// left(0)/right(1) path; each step repeats the sequential loop's exact
// (lo+hi)/2 recurrence, so the value is BITWISE what the sequential
// bisection would compute on that path.
__device__ inline double heap_midpoint(double lo, double hi, int k) {
  double x = 0.5 * (lo + hi);
  const int depth_below = 31 - __clz(k);
  for (int level = depth_below - 1; level >= 0; --level) {
    if ((k >> level) & 1) {
      lo = x;
    } else {
      hi = x;
    }
    x = 0.5 * (lo + hi);
  }
  return x;
}

// Speculative bisection for one search group.  All SW waves of the group
// execute this with identical control flow; `sub` is the wave's index
// within the group, `yb`/`fl` its SW-slot mailbox.  Returns the boundary
// indicator (-1/0/+1) and writes lambda* like wg_binary_search.
template <int D, typename F>
__device__ int spec_binary_search(double x_min, double x_max, double y_target,
                                  F eval, int sub, volatile double *yb,
                                  volatile double *fl, double *x_star) {
  constexpr int SW = (1 << D) - 1;
  const int lane = threadIdx.x & 63;

  // pre-phase: y(x_min) and y(x_max) evaluated concurrently (slots 0,1)
  double mine = 0.0;
  if (sub == 0) mine = eval(x_min);
  if (sub == 1) mine = eval(x_max);
  if (lane == 0 && sub < 2) mb_publish(&yb[sub], mine, &fl[sub], 1.0);
  if (!mb_wait(fl, 2, 1.0)) {  // watchdog: treat as unreachable target
    *x_star = x_min;
    return -1;
  }
  const double y0 = yb[0];
  const double y1 = yb[1];

  if (within_tolerance(y0, y_target, kTolerance)) {
    *x_star = x_min;
    return 0;
  }
  if (within_tolerance(y1, y_target, kTolerance)) {
    *x_star = x_max;
    return 0;
  }
  if (within_tolerance(y0, y1, kTolerance)) {
    if (y_target > fmax(y0, y1)) {
      *x_star = x_max;
      return +1;
    }
    *x_star = x_min;
    return -1;
  }
  const bool increasing = y0 < y1;
  if ((increasing && y_target < y0) || (!increasing && y_target > y0)) {
    *x_star = x_min;
    return -1;
  }
  if ((increasing && y_target > y1) || (!increasing && y_target < y1)) {
    *x_star = x_max;
    return +1;
  }

  double lo = x_min, hi = x_max, xs = x_min;
  double round_id = 2.0;
  int iters = 0;
  while (iters < kMaxIterations) {
    // evaluate the whole depth-D midpoint tree of [lo, hi]
    const double x_mine = heap_midpoint(lo, hi, sub + 1);
    const double y_mine = eval(x_mine);
    if (lane == 0) mb_publish(&yb[sub], y_mine, &fl[sub], round_id);
    if (!mb_wait(fl, SW, round_id)) {
      *x_star = x_min;
      return -1;
    }
    round_id += 1.0;

    // replay the sequential walk on the cached tree
    int k = 1;
    bool done = false;
    for (int level = 0; level < D && iters < kMaxIterations; ++level) {
      xs = 0.5 * (lo + hi);
      const double ys = yb[k - 1];
      ++iters;
      if (within_tolerance(ys, y_target, kTolerance)) {
        done = true;
        break;
      }
      if ((increasing && y_target < ys) || (!increasing && y_target > ys)) {
        hi = xs;
        k = 2 * k;
      } else {
        lo = xs;
        k = 2 * k + 1;
      }
    }
    if (done) break;
  }
  *x_star = xs;
  return 0;
}

// Mailbox LDS layout (doubles, after cum[max_k] + totals[THREADS]):
//   [0..6]   group-0 y slots      [7..13]  group-0 flags
//   [14..20] group-1 y slots      [21..27] group-1 flags
//   [28..31] results: lam0, ind0, lam1, ind1
//   [32..33] done flags per group
template <int D>
__device__ void solve_body_spec(const double *__restrict__ prob,
                                double *__restrict__ out, int n_problems,
                                int max_k) {
  constexpr int SW = (1 << D) - 1;
  constexpr int THREADS = 2 * SW * 64;
  const int pid = blockIdx.x;
  if (pid >= n_problems) return;
  const double *pr = prob + (size_t)pid * PROBLEM_FIELDS;
  double *res = out + (size_t)pid * RESULT_FIELDS;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int group = wave / SW;   // 0 = TTFT, 1 = ITL
  const int sub = wave % SW;
  const int lane = tid & 63;

  const Parms p = load_parms(pr);
  const int K = p.max_batch * (1 + kMaxQueueToBatchRatio);

  extern __shared__ double smem[];
  double *cum = smem;
  double *totals = smem + max_k;
  double *mail = totals + THREADS;
  if (tid < 34) mail[tid] = 0.0;

  build_cum<THREADS>(p, K, cum, totals);  // ends with __syncthreads()

  const double lam_min = serv_rate(p, 1) * kEpsilon;  // req/ms
  const double lam_max = serv_rate(p, p.max_batch) * (1.0 - kEpsilon);

  WgEval<64> ev{p, cum, nullptr, K};  // lane-indexed, barrier-free

  if (tid < RESULT_FIELDS) res[tid] = 0.0;

  volatile double *yb = mail + group * 14;
  volatile double *fl = yb + 7;
  volatile double *ex = mail + 28;
  volatile double *done = mail + 32;

  double lam_t = lam_max;
  int ind = 0;
  const double target = (group == 0) ? pr[P_TARGET_TTFT] : pr[P_TARGET_ITL];
  if (target > 0.0) {
    if (group == 0) {
      ind = spec_binary_search<D>(
          lam_min, lam_max, target,
          [&](double x) { return ev.eval_ttft(x); }, sub, yb, fl, &lam_t);
    } else {
      ind = spec_binary_search<D>(
          lam_min, lam_max, target,
          [&](double x) { return ev.eval_itl(x); }, sub, yb, fl, &lam_t);
    }
  }
  if (sub == 0 && lane == 0) {
    ex[group * 2 + 0] = lam_t;
    ex[group * 2 + 1] = (double)ind;
    __threadfence_block();
    done[group] = 1.0;
  }
  if (wave != 0) return;  // group-0 wave 0 finishes the tail alone

  if (!mb_wait(done, 2, 1.0)) return;  // watchdog: leave result infeasible
  if (ex[1] < 0.0 || ex[3] < 0.0) return;  // a target below reachable range

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

extern "C" __global__ void __launch_bounds__(384) wva_solve_kernel_384(
    const double *__restrict__ prob, double *__restrict__ out, int n_problems,
    int max_k) {
  solve_body_spec<2>(prob, out, n_problems, max_k);
}

extern "C" __global__ void __launch_bounds__(896) wva_solve_kernel_896(
    const double *__restrict__ prob, double *__restrict__ out, int n_problems,
    int max_k) {
  solve_body_spec<3>(prob, out, n_problems, max_k);
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
  // 64 KiB LDS per workgroup: cum[max_k] + totals[threads] + 64 slack
  const int lds_doubles = 64 * 1024 / 8;
  auto fits = [&](int thr) { return max_k + thr + 64 <= lds_doubles; };
  if (env != nullptr && std::strcmp(env, "896") == 0 && fits(896)) {
    threads = 896;
  } else if (env != nullptr && std::strcmp(env, "384") == 0 && fits(384)) {
    threads = 384;
  } else if (env != nullptr && std::strcmp(env, "256") == 0) {
    threads = 256;
  } else if (env != nullptr && std::strcmp(env, "128") == 0) {
    threads = 128;
  } else if (env != nullptr && std::strcmp(env, "64") == 0) {
    threads = 64;
  } else if (n_problems >= 512) {
    // Chip-filling launches are THROUGHPUT-bound: with the geometric
    // queue-tail (queue_core.h geo_tail) collapsing the sweep cost, the
    // dual kernel's 2 evaluations per round are the least redundant
    // work at every measured chip-filling point — B=4096 rows of
    // profiles/r02_kernel_geo.json: 0.33/0.44/1.04 ms at
    // K=704/2816/7700 vs spec-384's 0.75/1.01/1.21.
    threads = 128;
  } else {
    // Underfilled launches can't occupy 256 CUs, so the serial
    // bisection IS the runtime and the speculative-tree multisection
    // kernel (depth 2, 2 x 3 waves, 2 levels per round) wins every
    // measured point: 0.106/0.116/0.141 ms at B=192 K=704/2816/7700
    // (r01 best geometry: 0.175/0.277/0.360; profiles/r02_kernel_geo.json).
    threads = fits(384) ? 384 : 256;
  }
  const size_t smem = (size_t)(max_k + threads + 64) * sizeof(double);
  if (threads == 64) {
    hipLaunchKernelGGL(wva::wva_solve_kernel_64, dim3(n_problems), dim3(64), smem,
                       (hipStream_t)stream, prob, out, n_problems, max_k);
  } else if (threads == 128) {
    hipLaunchKernelGGL(wva::wva_solve_kernel_128, dim3(n_problems), dim3(128), smem,
                       (hipStream_t)stream, prob, out, n_problems, max_k);
  } else if (threads == 384) {
    hipLaunchKernelGGL(wva::wva_solve_kernel_384, dim3(n_problems), dim3(384), smem,
                       (hipStream_t)stream, prob, out, n_problems, max_k);
  } else if (threads == 896) {
    hipLaunchKernelGGL(wva::wva_solve_kernel_896, dim3(n_problems), dim3(896), smem,
                       (hipStream_t)stream, prob, out, n_problems, max_k);
  } else {
    hipLaunchKernelGGL(wva::wva_solve_kernel_256, dim3(n_problems), dim3(256), smem,
                       (hipStream_t)stream, prob, out, n_problems, max_k);
  }
}
