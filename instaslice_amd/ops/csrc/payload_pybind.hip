// Python bindings for the gfx950 payload kernels (payload_core.hpp).
// Built in-tree as instaslice_amd/ops/_payload*.so by build_native.py
// (hipcc --offload-arch=gfx950). Fails loudly at call time if no GPU.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "payload_core.hpp"

namespace py = pybind11;

PYBIND11_MODULE(_payload, m) {
  m.doc() = "gfx950 HIP payload kernels (vecadd / membw / busy / xcd census)";

  m.def("device_count", &payload::device_count,
        "Number of visible HIP devices (0 if no GPU/driver)");

  m.def(
      "run_vecadd",
      [](size_t n, int device) {
        py::gil_scoped_release rel;
        return payload::run_vecadd(n, device);
      },
      py::arg("n") = (1 << 20), py::arg("device") = 0,
      "c = a + b over n floats on `device`; returns max abs error (expect 0)");

  m.def(
      "run_membw",
      [](size_t bytes, int iters, int device) {
        py::gil_scoped_release rel;
        return payload::run_membw(bytes, iters, device);
      },
      py::arg("bytes") = (size_t{1} << 30), py::arg("iters") = 10,
      py::arg("device") = 0,
      "streaming-copy bandwidth probe; returns GB/s (read+write)");

  m.def(
      "run_busy",
      [](double ms, int device) {
        py::gil_scoped_release rel;
        payload::run_busy(ms, device);
      },
      py::arg("ms") = 100.0, py::arg("device") = 0,
      "occupy the device with a bounded spin for ~ms (sleep-pod payload)");

  m.def(
      "run_xcd_census",
      [](int device, int blocks) {
        py::gil_scoped_release rel;
        return payload::run_xcd_census(device, blocks);
      },
      py::arg("device") = 0, py::arg("blocks") = 4096,
      "per-XCD workgroup counts: exactly one nonzero in a CPX partition");

  m.def(
      "device_info",
      [](int device) {
        payload::DeviceInfo info;
        {
          py::gil_scoped_release rel;
          info = payload::get_device_info(device);
        }
        py::dict d;
        d["device"] = info.device;
        d["name"] = info.name;
        d["gcn_arch"] = info.gcn_arch;
        d["cu_count"] = info.cu_count;
        d["total_mem_gb"] = info.total_mem_gb;
        d["xcd_count_visible"] = info.xcd_count_visible;
        return d;
      },
      py::arg("device") = 0);
}
