// Python bindings for the partitiond device layer (see partitiond_core.hpp).
// Built in-tree as instaslice_amd/smi/_partitiond*.so by build_native.py.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "partitiond_core.hpp"

namespace py = pybind11;
using namespace partitiond;

PYBIND11_MODULE(_partitiond, m) {
  m.doc() = "C++ device layer over libamd_smi.so (MI355X partitioning)";

  static py::exception<SmiException> exc(m, "SmiNativeError");
  py::register_exception_translator([](std::exception_ptr p) {
    try {
      if (p) std::rethrow_exception(p);
    } catch (const SmiException& e) {
      // carry the amdsmi status code so Python can map EBUSY/PERM/UNSUPPORTED
      PyErr_SetObject(
          exc.ptr(),
          py::make_tuple(py::str(e.what()), py::int_(static_cast<int>(e.status)))
              .ptr());
    }
  });

  py::class_<ProcessorInfo>(m, "ProcessorInfo")
      .def_readonly("index", &ProcessorInfo::index)
      .def_readonly("uuid", &ProcessorInfo::uuid)
      .def_readonly("asic_name", &ProcessorInfo::asic_name)
      .def_readonly("asic_serial", &ProcessorInfo::asic_serial)
      .def_readonly("vram_total_mb", &ProcessorInfo::vram_total_mb)
      .def_readonly("bdf", &ProcessorInfo::bdf)
      .def_readonly("node_id", &ProcessorInfo::node_id)
      .def_readonly("partition_id", &ProcessorInfo::partition_id)
      .def_readonly("compute_partition", &ProcessorInfo::compute_partition)
      .def_readonly("memory_partition", &ProcessorInfo::memory_partition)
      .def_readonly("num_compute_units", &ProcessorInfo::num_compute_units)
      .def("__repr__", [](const ProcessorInfo& p) {
        return "<ProcessorInfo idx=" + std::to_string(p.index) + " uuid=" + p.uuid +
               " mode=" + p.compute_partition + "/" + p.memory_partition +
               " part_id=" + std::to_string(p.partition_id) + ">";
      });

  py::class_<ProfileInfo>(m, "ProfileInfo")
      .def_readonly("profile_type", &ProfileInfo::profile_type)
      .def_readonly("num_partitions", &ProfileInfo::num_partitions)
      .def_readonly("profile_index", &ProfileInfo::profile_index)
      .def_readonly("memory_caps", &ProfileInfo::memory_caps);

  py::class_<Partitiond::LinkInfo>(m, "LinkInfo")
      .def_readonly("src", &Partitiond::LinkInfo::src)
      .def_readonly("dst", &Partitiond::LinkInfo::dst)
      .def_readonly("hops", &Partitiond::LinkInfo::hops)
      .def_readonly("weight", &Partitiond::LinkInfo::weight)
      .def_readonly("type", &Partitiond::LinkInfo::type);

  py::class_<Metrics>(m, "Metrics")
      .def_readonly("gfx_activity_pct", &Metrics::gfx_activity_pct)
      .def_readonly("umc_activity_pct", &Metrics::umc_activity_pct)
      .def_readonly("vram_used_mb", &Metrics::vram_used_mb)
      .def_readonly("socket_power_w", &Metrics::socket_power_w);

  py::class_<Partitiond>(m, "Partitiond")
      .def(py::init<>())
      .def("init", &Partitiond::init, py::call_guard<py::gil_scoped_release>())
      .def("shutdown", &Partitiond::shutdown, py::call_guard<py::gil_scoped_release>())
      .def("enumerate", &Partitiond::enumerate, py::call_guard<py::gil_scoped_release>())
      .def("num_processors", &Partitiond::num_processors,
           py::call_guard<py::gil_scoped_release>())
      .def("get_compute_partition", &Partitiond::get_compute_partition,
           py::call_guard<py::gil_scoped_release>())
      .def("set_compute_partition", &Partitiond::set_compute_partition,
           py::call_guard<py::gil_scoped_release>())
      .def("get_memory_partition", &Partitiond::get_memory_partition,
           py::call_guard<py::gil_scoped_release>())
      .def("set_memory_partition", &Partitiond::set_memory_partition,
           py::call_guard<py::gil_scoped_release>())
      .def("get_profile_config", &Partitiond::get_profile_config,
           py::call_guard<py::gil_scoped_release>())
      .def("set_accelerator_profile", &Partitiond::set_accelerator_profile,
           py::call_guard<py::gil_scoped_release>())
      .def("get_current_profile", &Partitiond::get_current_profile,
           py::call_guard<py::gil_scoped_release>())
      .def("get_link_topology", &Partitiond::get_link_topology,
           py::call_guard<py::gil_scoped_release>())
      .def("get_metrics", &Partitiond::get_metrics,
           py::call_guard<py::gil_scoped_release>());

  // amdsmi status codes Python needs to classify failures
  m.attr("STATUS_BUSY") = static_cast<int>(AMDSMI_STATUS_BUSY);
  m.attr("STATUS_NO_PERM") = static_cast<int>(AMDSMI_STATUS_NO_PERM);
  m.attr("STATUS_NOT_SUPPORTED") = static_cast<int>(AMDSMI_STATUS_NOT_SUPPORTED);
  m.attr("STATUS_SETTING_UNAVAILABLE") =
      static_cast<int>(AMDSMI_STATUS_SETTING_UNAVAILABLE);
}
