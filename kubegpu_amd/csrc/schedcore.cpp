// schedcore — native scheduler hot path (pybind11).
//
// Exact C++ twin of kubegpu_amd/scheduler/xgmi.py: max-bottleneck ring
// (bitmask DP) + subset chooser with (ring bw, remaining-xGMI-edges,
// aggregate ring bw, lexicographic) ordering.  Iteration order and
// tie-breaking mirror the Python reference exactly; tests assert
// equivalence (tests/test_xgmi.py::test_fast_path_matches_python).
// p50 pod-schedule latency is a headline metric, and
// subset choice is the only super-linear piece of the schedule path.
//
// Build: g++ -O3 -shared -fPIC schedcore.cpp $(python -m pybind11 --includes)
//        -o ../_schedcore$(python3-config --extension-suffix)

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <limits>
#include <tuple>
#include <vector>

namespace py = pybind11;

namespace {

constexpr double kXgmiClassGbps = 100.0;  // xgmi.py XGMI_CLASS_GBPS
constexpr double kInfCap = 1e9;

struct Ring {
  double bottleneck;
  std::vector<int> order;
};

// bw is a flat n*n row-major matrix, already symmetrized by the caller
// (Python passes min(bw[i][j], bw[j][i])).
Ring best_ring(const std::vector<int>& sub, const double* bw, int n) {
  int k = (int)sub.size();
  if (k == 0) return {0.0, {}};
  if (k == 1) return {std::numeric_limits<double>::infinity(), {sub[0]}};
  auto B = [&](int a, int b) { return bw[a * n + b]; };
  if (k == 2) return {B(sub[0], sub[1]), {sub[0], sub[1]}};
  int start = sub[0];
  std::vector<int> rest(sub.begin() + 1, sub.end());
  int m = (int)rest.size();
  size_t states = (size_t)1 << m;
  std::vector<double> dp(states * m, -1.0);
  std::vector<int32_t> par(states * m, -1);
  for (int i = 0; i < m; ++i) dp[((size_t)1 << i) * m + i] = B(start, rest[i]);
  for (size_t mask = 1; mask < states; ++mask) {
    for (int last = 0; last < m; ++last) {
      if (!((mask >> last) & 1)) continue;
      double cur = dp[mask * m + last];
      if (cur < 0.0) continue;
      for (int nxt = 0; nxt < m; ++nxt) {
        if ((mask >> nxt) & 1) continue;
        double val = std::min(cur, B(rest[last], rest[nxt]));
        size_t key = (mask | ((size_t)1 << nxt)) * m + nxt;
        if (val > dp[key]) {
          dp[key] = val;
          par[key] = (int32_t)(mask * m + last);
        }
      }
    }
  }
  size_t full = states - 1;
  double best_val = -1.0;
  int best_last = -1;
  for (int last = 0; last < m; ++last) {
    double cur = dp[full * m + last];
    if (cur < 0.0) continue;
    double closed = std::min(cur, B(rest[last], start));
    if (closed > best_val) {
      best_val = closed;
      best_last = last;
    }
  }
  std::vector<int> order;
  int64_t key = (int64_t)(full * m + best_last);
  while (par[key] >= 0) {
    order.push_back(rest[key % m]);
    key = par[key];
  }
  order.push_back(rest[key % m]);
  order.push_back(start);
  std::reverse(order.begin(), order.end());
  return {std::max(best_val, 0.0), order};
}

int xgmi_edges(const std::vector<int>& gpus, const double* bw, int n) {
  int count = 0;
  for (size_t a = 0; a < gpus.size(); ++a)
    for (size_t b = a + 1; b < gpus.size(); ++b)
      if (bw[gpus[a] * n + gpus[b]] >= kXgmiClassGbps) ++count;
  return count;
}

std::tuple<double, int, double> score_subset(const std::vector<int>& sub,
                                             const std::vector<int>& free,
                                             const double* bw, int n) {
  Ring r = best_ring(sub, bw, n);
  std::vector<bool> in_sub(n, false);
  for (int g : sub) in_sub[g] = true;
  std::vector<int> remaining;
  for (int g : free)
    if (!in_sub[g]) remaining.push_back(g);
  int frag = xgmi_edges(remaining, bw, n);
  double agg = 0.0;
  if (r.order.size() >= 3) {
    for (size_t i = 0; i < r.order.size(); ++i)
      agg += bw[r.order[i] * n + r.order[(i + 1) % r.order.size()]];
  } else if (r.order.size() == 2) {
    agg = bw[r.order[0] * n + r.order[1]];
  }
  double cap = std::isinf(r.bottleneck) ? kInfCap : r.bottleneck;
  return {cap, frag, agg};
}

// Candidates are positions 0..n-1 (the Python wrapper maps real GPU
// indices); lexicographic combination order + strictly-greater keeps the
// same winner as itertools.combinations in xgmi.py.  `must` (positions)
// constrains the search to supersets of that set — the kubelet
// GetPreferredAllocation contract (must_include_deviceIDs).
std::vector<int> choose_best_subset(int n, int k, const std::vector<double>& bw_flat,
                                    const std::vector<int>& must) {
  const double* bw = bw_flat.data();
  std::vector<int> free(n);
  for (int i = 0; i < n; ++i) free[i] = i;
  if (k <= 0 || k > n || (int)must.size() > k) return {};
  if (k == n) return free;
  uint64_t must_mask = 0;
  for (int g : must) {
    if (g < 0 || g >= n) return {};
    must_mask |= (uint64_t)1 << g;
  }
  std::vector<int> comb(k);
  for (int i = 0; i < k; ++i) comb[i] = i;
  std::tuple<double, int, double> best{-1.0, -1, -1.0};
  std::vector<int> best_sub;
  while (true) {
    uint64_t mask = 0;
    for (int g : comb) mask |= (uint64_t)1 << g;
    if ((mask & must_mask) == must_mask) {
      auto s = score_subset(comb, free, bw, n);
      if (s > best) {
        best = s;
        best_sub = comb;
      }
    }
    // next lexicographic combination
    int i = k - 1;
    while (i >= 0 && comb[i] == n - k + i) --i;
    if (i < 0) break;
    ++comb[i];
    for (int j = i + 1; j < k; ++j) comb[j] = comb[j - 1] + 1;
  }
  return best_sub;
}

std::pair<double, std::vector<int>> best_ring_py(int n, const std::vector<int>& sub,
                                                 const std::vector<double>& bw_flat) {
  Ring r = best_ring(sub, bw_flat.data(), n);
  double v = std::isinf(r.bottleneck) ? kInfCap : r.bottleneck;
  return {v, r.order};
}

}  // namespace

PYBIND11_MODULE(_schedcore, m) {
  m.doc() = "native xGMI subset scorer (twin of kubegpu_amd.scheduler.xgmi)";
  m.def("choose_best_subset", &choose_best_subset, py::arg("n"), py::arg("k"),
        py::arg("bw_flat"), py::arg("must") = std::vector<int>());
  m.def("best_ring", &best_ring_py, py::arg("n"), py::arg("subset"),
        py::arg("bw_flat"));
}
