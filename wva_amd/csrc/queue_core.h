// Shared scalar math for the batched state-dependent M/M/1/K allocation
// sizing kernel.  Mirrors wva_amd/analyzer (which itself re-implements the
// behavior of /root/reference/pkg/analyzer) in double precision:
//   - service rates  mu(b) = b / (prefill(b) + numDecode * decode(b))
//   - product-form state probabilities in log space (softmax-normalized)
//   - monotone bisection with boundary-region classification
//   - Size(): max rate meeting TTFT/ITL targets; TPS as lambdaMax*0.9
//
// Used by both the CPU (OpenMP) path and the gfx950 HIP kernel.
#pragma once

#include <cmath>

#if defined(__HIPCC__)
#define WVA_HD __host__ __device__ inline
#else
#define WVA_HD inline
#endif

namespace wva {

constexpr double kEpsilon = 0.001;           // rate-range disturbance
constexpr double kStabilityFraction = 0.1;   // TPS backoff fraction
constexpr double kTolerance = 1e-6;          // bisection relative tolerance
constexpr int kMaxIterations = 100;          // bisection iteration cap
constexpr int kMaxQueueToBatchRatio = 10;    // K = (1 + ratio) * N

// Problem layout: one row of 12 doubles.
enum ProblemField {
  P_ALPHA = 0,    // decode base (ms)
  P_BETA,         // decode slope (ms per batch unit)
  P_GAMMA,        // prefill base (ms)
  P_DELTA,        // prefill slope (ms per token per batch unit)
  P_IN_TOKENS,    // average input tokens
  P_OUT_TOKENS,   // average output tokens (K >= 1)
  P_MAX_BATCH,    // N >= 1
  P_TARGET_TTFT,  // ms; 0 disables
  P_TARGET_ITL,   // ms; 0 disables
  P_TARGET_TPS,   // tokens/s; 0 disables
  P_TOTAL_RATE,   // req/s (> 0; zero-load handled by the host)
  P_MIN_REPLICAS, // minimum replica count
  PROBLEM_FIELDS
};

// Result layout: one row of 6 doubles.
enum ResultField {
  R_FEASIBLE = 0, // 1.0 feasible / 0.0 infeasible
  R_REPLICAS,     // number of replicas
  R_RATE_STAR,    // max per-replica rate meeting targets (req/s)
  R_ITL,          // predicted decode time per token at the final rate (ms)
  R_TTFT,         // predicted wait + prefill at the final rate (ms)
  R_RHO,          // avg-in-service / max-batch, clamped to [0,1]
  RESULT_FIELDS
};

struct Parms {
  double alpha, beta, gamma, delta;
  double in_tokens;
  int out_tokens;
  int max_batch;   // N
  int num_decode;  // out_tokens-1, or 1 for decode-only single-token
};

WVA_HD double prefill_time(const Parms &p, double batch) {
  return p.in_tokens == 0.0 ? 0.0 : p.gamma + p.delta * p.in_tokens * batch;
}

WVA_HD double decode_time(const Parms &p, double batch) {
  return p.alpha + p.beta * batch;
}

// state-dependent service rate for batch size b in [1, N] (req/ms)
WVA_HD double serv_rate(const Parms &p, int b) {
  return (double)b / (prefill_time(p, (double)b) + p.num_decode * decode_time(p, (double)b));
}

// log(mu(state n)) for the birth-death recursion, n in [0, K-1]
WVA_HD double log_mu(const Parms &p, int n) {
  int b = n + 1;
  if (b > p.max_batch) b = p.max_batch;
  return log(serv_rate(p, b));
}

WVA_HD bool within_tolerance(double x, double value, double tolerance) {
  if (x == value) return true;
  if (value == 0.0 || tolerance < 0.0) return false;
  return fabs((x - value) / value) <= tolerance;
}

// effective concurrency: solve the service-time identity for n, clamped
WVA_HD double effective_concurrency(const Parms &p, double avg_serv_time) {
  double tokens = (double)(p.out_tokens - 1);
  double numer = avg_serv_time - (p.gamma + p.alpha * tokens);
  double denom = p.delta * p.in_tokens + p.beta * tokens;
  if (denom == 0.0) return numer > 0.0 ? (double)p.max_batch : 0.0;
  double n = numer / denom;
  if (n < 0.0) n = 0.0;
  if (n > (double)p.max_batch) n = (double)p.max_batch;
  return n;
}

struct Stats {
  double throughput;  // req/ms
  double wait;        // ms
  double serv;        // ms
  double n_serv;      // avg requests in service
};

// Mode of the log-probability curve logp[n] = n*log(lam) - cum[n-1].
// The increment logp[n+1]-logp[n] = log(lam) - log(mu(n)) is nonincreasing
// in n because mu(n) = serv_rate(min(n+1, N)) is nondecreasing, so logp is
// concave and its maximum sits where the increment first turns <= 0:
// an O(log N) search over the monotone service-rate curve instead of an
// O(K) max sweep.  Returns the state index n* in [0, K].
// serv_rate(b) < lam  <=>  b < lam * T(b)  (service time T(b) > 0):
// a multiply instead of the fp64 divide inside the mode bisection
WVA_HD bool rate_below(const Parms &p, int b, double lam) {
  double t = prefill_time(p, (double)b) + p.num_decode * decode_time(p, (double)b);
  return (double)b < lam * t;
}

WVA_HD int log_mode_state(const Parms &p, int K, double lam) {
  // lam >= mu(N)  <=>  N <= lam * T(N): all increments > 0, mode at K
  double tN = prefill_time(p, (double)p.max_batch) +
              p.num_decode * decode_time(p, (double)p.max_batch);
  if ((double)p.max_batch <= lam * tN) return K;
  // lam <= mu(1)  <=>  lam * T(1) <= 1: mode at 0
  double t1 = prefill_time(p, 1.0) + p.num_decode * decode_time(p, 1.0);
  if (lam * t1 <= 1.0) return 0;
  // smallest b in [1, N] with serv_rate(b) >= lam; mode = b - 1
  int lo = 1, hi = p.max_batch;
  while (lo < hi) {
    int mid = (lo + hi) / 2;
    if (rate_below(p, mid, lam)) {
      lo = mid + 1;
    } else {
      hi = mid;
    }
  }
  return lo - 1;
}

WVA_HD double log_p(const double *cum, double loglam, int n) {
  return (n == 0) ? 0.0 : n * loglam - cum[n - 1];
}

// States more than kLogCutoff nats below the mode contribute < ~1e-16 to
// any normalized sum (at most K * e^-45 relative mass); since logp is
// concave, the significant window [lo, hi] around the mode is found by
// two O(log K) bisections instead of sweeping all K+1 states.
constexpr double kLogCutoff = 45.0;

WVA_HD void state_window(const double *cum, double loglam, int K, int n_star,
                         double m, int *lo_out, int *hi_out) {
  const double thresh = m - kLogCutoff;
  // left edge: logp nondecreasing on [0, n_star]; first n with logp >= thresh
  int lo = 0, hi = n_star;
  while (lo < hi) {
    int mid = (lo + hi) / 2;
    if (log_p(cum, loglam, mid) < thresh) {
      lo = mid + 1;
    } else {
      hi = mid;
    }
  }
  *lo_out = lo;
  // right edge: logp nonincreasing on [n_star, K]; last n with logp >= thresh
  lo = n_star;
  hi = K;
  while (lo < hi) {
    int mid = (lo + hi + 1) / 2;
    if (log_p(cum, loglam, mid) < thresh) {
      hi = mid - 1;
    } else {
      lo = mid;
    }
  }
  *hi_out = lo;
}

// Closed-form geometric tail of the state sums (the "queue region").
//
// For n >= N-1 the service rate is constant (mu(n) = serv_rate(N)), so
// consecutive log-probabilities differ by the constant d = log(lam) -
// log(mu_N) and the normalized probabilities e^{logp(n)-m} form a
// geometric sequence.  Summing states [g_start, n_hi] (g_start > N, so
// none of them contribute to the <=N partial sums) in closed form
// replaces up to 10*N per-state exp() calls — the dominant fp64 cost of
// an evaluation — with a handful of exp/expm1.
//
//   S_geo  = p_a * G0,            G0 = sum_{j=0..M} r^j = expm1((M+1)d)/expm1(d)
//   Ni_geo = p_a * (a*G0 + G1),   G1 = sum_{j=0..M} j r^j
//                                    = r (1 - (M+1) r^M + M r^{M+1}) / (1-r)^2
//
// Near r == 1 the G1 numerator cancels; |d|*(M+1) < 0.01 falls back to
// the term-wise loop (a one-or-two-iterate sliver of the bisection).
struct GeoTail {
  double S, Ni, eK;
  bool used;
};

WVA_HD GeoTail geo_tail(const double *cum, double loglam, double d, double m,
                        int g_start, int n_hi, int K) {
  GeoTail t{0.0, 0.0, 0.0, false};
  const int M = n_hi - g_start;
  if (M < 8 || fabs(d) * (double)(M + 1) < 0.01) return t;  // loop is fine/safer
  const double x_a = log_p(cum, loglam, g_start) - m;
  const double p_a = exp(x_a);
  const double e1 = expm1(d);
  const double G0 = expm1((double)(M + 1) * d) / e1;
  const double r = e1 + 1.0;
  const double rM = exp((double)M * d);
  const double G1 =
      r * (1.0 - (double)(M + 1) * rM + (double)M * rM * r) / (e1 * e1);
  t.S = p_a * G0;
  t.Ni = p_a * ((double)g_start * G0 + G1);
  if (n_hi == K) t.eK = p_a * exp((double)(K - g_start) * d);
  t.used = true;
  return t;
}

// Scalar model evaluation at arrival rate lam (req/ms) given the inclusive
// cumulative sum cum[n] = sum_{i<=n} log_mu(i), n in [0, K-1].
WVA_HD Stats eval_model(const Parms &p, const double *cum, int K, double lam) {
  double loglam = log(lam);
  // max of logp over n = 0..K via the concavity closed form
  int n_star = log_mode_state(p, K, lam);
  double m = log_p(cum, loglam, n_star);
  if (m < 0.0) m = 0.0;  // logp(0) = 0 participates in the max
  int n_lo, n_hi;
  state_window(cum, loglam, K, n_star, m, &n_lo, &n_hi);
  int num = p.max_batch;  // serv_rate array length
  double S = 0.0, Ni = 0.0, Snum = 0.0, Ninum = 0.0, eK = 0.0;
  // geometric tail over the constant-rate queue region (states > N)
  const int g_start = (num + 1 > n_lo) ? num + 1 : n_lo;
  GeoTail tail{0.0, 0.0, 0.0, false};
  int loop_hi = n_hi;
  if (g_start + 8 <= n_hi) {  // also keeps cum[g_start] in bounds
    const double d = loglam - (cum[g_start] - cum[g_start - 1]);
    tail = geo_tail(cum, loglam, d, m, g_start, n_hi, K);
    if (tail.used) loop_hi = g_start - 1;
  }
  for (int n = n_lo; n <= loop_hi; ++n) {
    double e = exp(log_p(cum, loglam, n) - m);
    S += e;
    Ni += n * e;
    if (n <= num) {
      Snum += e;
      Ninum += n * e;
    }
    if (n == K) eK = e;
  }
  S += tail.S;
  Ni += tail.Ni;
  eK += tail.eK;
  Stats st;
  st.throughput = lam * (1.0 - eK / S);
  double n_sys = Ni / S;
  st.n_serv = Ninum / S + (1.0 - Snum / S) * num;
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

WVA_HD double eval_ttft_of(const Parms &p, const Stats &st) {
  double effc = effective_concurrency(p, st.serv);
  return st.wait + prefill_time(p, effc);
}

WVA_HD double eval_itl_of(const Parms &p, const Stats &st) {
  double effc = effective_concurrency(p, st.serv);
  return decode_time(p, effc);
}

}  // namespace wva
