// C++ DP search core (pybind11 module _galvatron_dp_core).
//
// Reference role: galvatron/csrc/dp_core.cpp:24-122 (the load-bearing
// O(L*M*S^2) knapsack recurrence behind the strategy search). Interface
// re-designed: fills caller-allocated numpy f/mark tables; back-trace and
// the per-vocab-tp variants live in Python (search/dp.py).
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cmath>
#include <cstdint>
#include <limits>
#include <vector>

namespace py = pybind11;

void dynamic_programming_core(int layer_num, int max_mem, int strategy_num,
                              py::array_t<int32_t> v_data_a,
                              py::array_t<double> intra_a,
                              py::array_t<double> inter_a,
                              py::array_t<double> f_a,
                              py::array_t<int16_t> mark_a) {
  const auto v_data = v_data_a.unchecked<2>();   // [L, S]
  const auto intra = intra_a.unchecked<2>();     // [L, S]
  const auto inter = inter_a.unchecked<3>();     // [L, S, S] (si -> s)
  auto f = f_a.mutable_unchecked<2>();           // [M, S]
  auto mark = mark_a.mutable_unchecked<3>();     // [L, M, S]
  // accessors above hold raw buffers; the O(L*M*S^2) loops below never
  // touch the Python API, so drop the GIL and let the engine's
  // parallel_search thread pool overlap DP tasks
  py::gil_scoped_release release;
  const double INF = std::numeric_limits<double>::infinity();

  std::vector<double> prev(max_mem * strategy_num, 0.0);
  std::vector<double> cur(max_mem * strategy_num, INF);
  for (int v = 0; v < max_mem; ++v)
    for (int s = 0; s < strategy_num; ++s) prev[v * strategy_num + s] = 0.0;

  for (int i = 0; i < layer_num; ++i) {
    for (int v = 0; v < max_mem; ++v)
      for (int s = 0; s < strategy_num; ++s)
        cur[v * strategy_num + s] = INF;
    for (int s = 0; s < strategy_num; ++s) {
      const int vd = v_data(i, s);
      if (vd >= max_mem) continue;
      const double ic = intra(i, s);
      for (int v = vd; v < max_mem; ++v) {
        const double* pv = &prev[(v - vd) * strategy_num];
        double best = INF;
        int best_si = -1;
        for (int si = 0; si < strategy_num; ++si) {
          const double c = pv[si] + inter(i, si, s);
          if (c < best) {
            best = c;
            best_si = si;
          }
        }
        cur[v * strategy_num + s] = best + ic;
        mark(i, v, s) = (int16_t)best_si;
      }
    }
    prev.swap(cur);
  }
  for (int v = 0; v < max_mem; ++v)
    for (int s = 0; s < strategy_num; ++s)
      f(v, s) = prev[v * strategy_num + s];
}

PYBIND11_MODULE(_galvatron_dp_core, m) {
  m.doc() = "layer-wise hybrid-parallel DP search core";
  m.def("dynamic_programming_core", &dynamic_programming_core,
        py::arg("layer_num"), py::arg("max_mem"), py::arg("strategy_num"),
        py::arg("v_data"), py::arg("intra"), py::arg("inter"), py::arg("f"),
        py::arg("mark"));
}
