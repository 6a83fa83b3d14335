// Native (C++) request router: the reference's ext-proc scheduler is native
// Go; this is the MI355X build's native equivalent of the same hot path —
// the filter-tree decision over live engine gauges
// (`pkg/ext-proc/scheduling/filter.go` semantics, kept EXACTLY equal to the
// Python tree in ../filters.py; tests/test_native_router.py proves
// survivor-set equivalence on random fleets).
//
// Design: the router owns the slow-changing engine state (names, adapter
// sets, slot budgets) and reads the fast-changing gauges DIRECTLY from the
// engine pool's lock-free shared-memory array (serving/pool.py slot layout)
// through a raw pointer — per-request cost is the tree walk only (~1 µs at
// 200 engines vs ~105 µs for the Python tree).  Gauge mode falls back to
// per-engine stored values when no shared buffer is attached (remote-scrape
// mode, tests).
//
// All routing logic lives in router_core.h (pure C++) so the sanitizer
// harness (`make test-native-debug`) runs the production code under
// ASan/UBSan; this file is only the pybind11 binding.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "router_core.h"

namespace py = pybind11;

namespace {

class NativeRouter : public lig::RouterCore {
 public:
  explicit NativeRouter(uint64_t seed) : lig::RouterCore(seed) {}

  void attach_shm_py(py::buffer buf, int slots_per_engine, int slot_waiting,
                     int slot_kv, int slot_draining) {
    py::buffer_info info = buf.request();
    if (info.itemsize != sizeof(double))
      throw std::runtime_error("shared gauge buffer must be doubles");
    attach_shm(static_cast<const double*>(info.ptr),
               static_cast<size_t>(info.size), slots_per_engine, slot_waiting,
               slot_kv, slot_draining);
    shm_holder_ = py::reinterpret_borrow<py::object>(buf);  // keep alive
  }

 private:
  py::object shm_holder_;
};

}  // namespace

PYBIND11_MODULE(_router, m) {
  m.doc() = "native filter-tree router (reference scheduling/ semantics)";
  py::class_<NativeRouter>(m, "NativeRouter")
      .def(py::init<uint64_t>(), py::arg("seed") = 0)
      .def("attach_shm", &NativeRouter::attach_shm_py, py::arg("buffer"),
           py::arg("slots_per_engine"), py::arg("slot_waiting"),
           py::arg("slot_kv"), py::arg("slot_draining"))
      .def("set_thresholds", &NativeRouter::set_thresholds)
      .def("upsert", &NativeRouter::upsert, py::arg("name"),
           py::arg("pool") = "", py::arg("slot_base") = -1,
           py::arg("adapters") = std::vector<std::string>{},
           py::arg("max_active") = 0)
      .def("set_gauges", &NativeRouter::set_gauges, py::arg("name"),
           py::arg("waiting"), py::arg("running"), py::arg("kv"),
           py::arg("draining") = false)
      .def("remove", &NativeRouter::remove)
      .def("size", &NativeRouter::size)
      .def("survivors", &NativeRouter::survivors, py::arg("resolved"),
           py::arg("critical"), py::arg("pool_ref") = "")
      .def("schedule", &NativeRouter::schedule, py::arg("resolved"),
           py::arg("critical"), py::arg("pool_ref") = "");
}
