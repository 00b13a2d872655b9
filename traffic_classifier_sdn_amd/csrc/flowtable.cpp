// pybind11 wrapper around the pure-C++ flow table core (flowtable_core.h):
// numpy views of the feature matrix / counter snapshots for the GPU predict
// path, plus the streaming ingestion entry points.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "flowtable_core.h"

namespace py = pybind11;
using tcsdn::NativeFlowTable;

namespace {

py::array_t<float> feature_matrix(const NativeFlowTable& t) {
  const size_t n = t.size();
  py::array_t<float> out({(py::ssize_t)n, (py::ssize_t)12});
  float* o = out.mutable_data();
  for (size_t i = 0; i < n; ++i) {
    const double* s = t.row(i);
    for (int j = 0; j < 12; ++j) o[i * 12 + j] = (float)s[tcsdn::kFeatureCols[j]];
  }
  return out;
}

py::tuple counters_snapshot(const NativeFlowTable& t) {
  using namespace tcsdn;
  const size_t n = t.size();
  py::array_t<double> cur({(py::ssize_t)n, (py::ssize_t)4});
  py::array_t<double> prev({(py::ssize_t)n, (py::ssize_t)4});
  py::array_t<double> times({(py::ssize_t)n, (py::ssize_t)6});
  double* c = cur.mutable_data();
  double* p = prev.mutable_data();
  double* tm = times.mutable_data();
  for (size_t i = 0; i < n; ++i) {
    const double* s = t.row(i);
    c[i * 4 + 0] = s[F_PKTS]; c[i * 4 + 1] = s[F_BYTES];
    c[i * 4 + 2] = s[R_PKTS]; c[i * 4 + 3] = s[R_BYTES];
    p[i * 4 + 0] = s[F_PREV_PKTS]; p[i * 4 + 1] = s[F_PREV_BYTES];
    p[i * 4 + 2] = s[R_PREV_PKTS]; p[i * 4 + 3] = s[R_PREV_BYTES];
    tm[i * 6 + 0] = s[F_LAST_TIME]; tm[i * 6 + 1] = s[F_PREV_TIME];
    tm[i * 6 + 2] = s[R_LAST_TIME]; tm[i * 6 + 3] = s[R_PREV_TIME];
    tm[i * 6 + 4] = s[TIME_START]; tm[i * 6 + 5] = 0.0;
  }
  return py::make_tuple(cur, prev, times);
}

py::list statuses(const NativeFlowTable& t) {
  using namespace tcsdn;
  py::list out;
  for (size_t i = 0; i < t.size(); ++i) {
    const double* s = t.row(i);
    out.append(py::make_tuple(s[F_ACTIVE] != 0.0 ? "ACTIVE" : "INACTIVE",
                              s[R_ACTIVE] != 0.0 ? "ACTIVE" : "INACTIVE"));
  }
  return out;
}

py::list metas(const NativeFlowTable& t) {
  py::list out;
  for (const tcsdn::Meta& m : t.metas())
    out.append(py::make_tuple(m.datapath, m.inport, m.ethsrc, m.ethdst, m.outport));
  return out;
}

}  // namespace

PYBIND11_MODULE(_tcsdn_native, m) {
  m.doc() = "native flow table + telemetry parser";
  py::class_<NativeFlowTable>(m, "NativeFlowTable")
      .def(py::init<>())
      .def("__len__", &NativeFlowTable::size)
      .def("update", &NativeFlowTable::update, py::arg("time"),
           py::arg("datapath"), py::arg("inport"), py::arg("ethsrc"),
           py::arg("ethdst"), py::arg("outport"), py::arg("packets"),
           py::arg("bytes"))
      .def("feed_line",
           [](NativeFlowTable& t, std::string_view line) { return t.feed_line(line); })
      .def("feed_buffer",
           [](NativeFlowTable& t, std::string_view buf) { return t.feed_buffer(buf); })
      .def("feature_matrix", &feature_matrix)
      .def("counters_snapshot", &counters_snapshot)
      .def("statuses", &statuses)
      .def("metas", &metas)
      .def_readonly("records", &NativeFlowTable::records)
      .def_readonly("bad_lines", &NativeFlowTable::bad_lines);
}
