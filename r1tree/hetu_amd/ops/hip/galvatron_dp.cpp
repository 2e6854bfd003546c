// Native dynamic-programming core for the Galvatron-style strategy search.
//
// Reference parity: tools/Galvatron/csrc/dp_core.cpp:22-94 — per-layer
// strategy choice under a memory budget: minimize sum(time[l][s_l]) s.t.
// sum(mem[l][s_l]) <= cap, solved as a DP over quantized memory with
// backtracking.  Used here to pick per-layer recompute/strategy variants on
// 288 GB MI355X budgets.
#include <torch/extension.h>

#include <cmath>
#include <limits>
#include <vector>

// times/mems: [L][S] flattened; mem quantized to `buckets` levels of size
// cap/buckets. Returns (best_time, choices[L]); best_time = inf if
// infeasible.
std::pair<double, std::vector<int64_t>> galvatron_dp(
    std::vector<double> times, std::vector<double> mems, int64_t L,
    int64_t S, double cap, int64_t buckets) {
  const double INF = std::numeric_limits<double>::infinity();
  const double unit = cap / (double)buckets;
  // f[l][b] = min time for layers [0, l) using <= b memory units
  std::vector<double> f((L + 1) * (buckets + 1), INF);
  std::vector<int16_t> choice(L * (buckets + 1), -1);
  for (int64_t b = 0; b <= buckets; ++b) f[b] = 0.0;
  for (int64_t l = 0; l < L; ++l) {
    for (int64_t b = 0; b <= buckets; ++b) {
      double best = INF;
      int16_t arg = -1;
      for (int64_t s = 0; s < S; ++s) {
        double m = mems[l * S + s];
        int64_t mu = (int64_t)std::ceil(m / unit);
        if (mu > b) continue;
        double prev = f[l * (buckets + 1) + (b - mu)];
        if (prev == INF) continue;
        double t = prev + times[l * S + s];
        if (t < best) { best = t; arg = (int16_t)s; }
      }
      f[(l + 1) * (buckets + 1) + b] = best;
      choice[l * (buckets + 1) + b] = arg;
    }
  }
  std::vector<int64_t> out(L, -1);
  double best = f[L * (buckets + 1) + buckets];
  if (best < INF) {
    int64_t b = buckets;
    for (int64_t l = L - 1; l >= 0; --l) {
      int16_t s = choice[l * (buckets + 1) + b];
      out[l] = s;
      double m = mems[l * S + s];
      b -= (int64_t)std::ceil(m / unit);
    }
  }
  return {best, out};
}
