// pybind11 bindings for the MI355X HIP kernel library -> gpudpf._hip.
// Tensor memory is owned by torch on the python side; this module receives
// raw device pointers (as integers) plus the current HIP stream, launches
// kernels, and manages the small AES table buffer per device.

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>

#include <map>
#include <mutex>
#include <stdexcept>
#include <string>

#include "dpf_core.h"
#include "dpf_hip_api.h"

namespace py = pybind11;

namespace {

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string("HIP error: ") +               \
                               hipGetErrorString(_e));                    \
  } while (0)

// Per-device cached AES table buffer (5*256 u32: te0..te3, sbox).
std::mutex g_aes_mu;
std::map<int, void*> g_aes_tables;

std::uintptr_t ensure_aes_tables(int device) {
  std::lock_guard<std::mutex> lock(g_aes_mu);
  auto it = g_aes_tables.find(device);
  if (it != g_aes_tables.end()) return reinterpret_cast<std::uintptr_t>(it->second);
  gpudpf::u32 host[5 * 256];
  gpudpf::aes128_tables(host, host + 256, host + 512, host + 768, host + 1024);
  int prev = 0;
  HIP_CHECK(hipGetDevice(&prev));
  HIP_CHECK(hipSetDevice(device));
  void* dev = nullptr;
  HIP_CHECK(hipMalloc(&dev, sizeof(host)));
  HIP_CHECK(hipMemcpy(dev, host, sizeof(host), hipMemcpyHostToDevice));
  HIP_CHECK(hipSetDevice(prev));
  g_aes_tables[device] = dev;
  return reinterpret_cast<std::uintptr_t>(dev);
}

int device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return 0;
  return n;
}

}  // namespace

PYBIND11_MODULE(_hip, m) {
  m.doc() = "gpudpf MI355X HIP kernels (gfx950)";
  m.def("eval_fused", &gpudpf_hip::launch_fused, py::arg("keys"),
        py::arg("table"), py::arg("out"), py::arg("aes_tabs"), py::arg("batch"),
        py::arg("n"), py::arg("depth"), py::arg("zlog"), py::arg("prf"),
        py::arg("stream"),
        py::call_guard<py::gil_scoped_release>());
  m.def("eval_expand", &gpudpf_hip::launch_expand, py::arg("keys"),
        py::arg("out"), py::arg("aes_tabs"), py::arg("batch"), py::arg("n"),
        py::arg("depth"), py::arg("zlog"), py::arg("prf"), py::arg("stream"),
        py::call_guard<py::gil_scoped_release>());
  m.def("eval_bfs", &gpudpf_hip::launch_bfs, py::arg("keys"), py::arg("out"),
        py::arg("aes_tabs"), py::arg("batch"), py::arg("n"), py::arg("depth"),
        py::arg("prf"), py::arg("stream"),
        py::call_guard<py::gil_scoped_release>());
  m.def("eval_coop", &gpudpf_hip::launch_coop, py::arg("keys"),
        py::arg("table"), py::arg("out"), py::arg("aes_tabs"), py::arg("n"),
        py::arg("depth"), py::arg("zlog"), py::arg("prf"), py::arg("fused"),
        py::arg("stream"), py::call_guard<py::gil_scoped_release>());
  m.def("eval_naive", &gpudpf_hip::launch_naive, py::arg("keys"),
        py::arg("out"), py::arg("aes_tabs"), py::arg("batch"), py::arg("n"),
        py::arg("depth"), py::arg("prf"), py::arg("stream"),
        py::call_guard<py::gil_scoped_release>());
  m.def("gemm128", &gpudpf_hip::launch_gemm128, py::arg("a"), py::arg("bt"),
        py::arg("c"), py::arg("partials"), py::arg("m"), py::arg("n"),
        py::arg("k"), py::arg("stream"),
        py::call_guard<py::gil_scoped_release>());
  m.def("gemm128_ksplit", &gpudpf_hip::gemm128_ksplit);
  m.def("prf_sol", &gpudpf_hip::launch_prf_sol, py::arg("aes_tabs"),
        py::arg("out"), py::arg("blocks"), py::arg("iters"), py::arg("prf"),
        py::arg("stream"), py::call_guard<py::gil_scoped_release>());
  m.def("digits", &gpudpf_hip::launch_digits, py::arg("inp"), py::arg("out"),
        py::arg("count"), py::arg("stream"),
        py::call_guard<py::gil_scoped_release>());
  m.def("gemm_u32_mfma", &gpudpf_hip::launch_gemm_u32_mfma, py::arg("da"),
        py::arg("dbt"), py::arg("c"), py::arg("m"), py::arg("n"), py::arg("k"),
        py::arg("stream"), py::call_guard<py::gil_scoped_release>());
  m.def("gemm_u32_stream", &gpudpf_hip::launch_gemm_u32_stream, py::arg("a"),
        py::arg("b"), py::arg("c"), py::arg("batch"), py::arg("k"),
        py::arg("n"), py::arg("stream"),
        py::call_guard<py::gil_scoped_release>());
  m.def("probe_alu", &gpudpf_hip::launch_probe_alu, py::arg("a"), py::arg("b"),
        py::arg("add_out"), py::arg("mul_out"), py::arg("count"),
        py::arg("stream"), py::call_guard<py::gil_scoped_release>());
  m.def("probe_prf", &gpudpf_hip::launch_probe_prf, py::arg("seeds"),
        py::arg("aes_tabs"), py::arg("pair0"), py::arg("pair1"),
        py::arg("single0"), py::arg("single1"), py::arg("low0"),
        py::arg("low1"), py::arg("count"), py::arg("prf"), py::arg("stream"),
        py::call_guard<py::gil_scoped_release>());
  m.def("ensure_aes_tables", &ensure_aes_tables, py::arg("device"));
  m.def("device_count", &device_count);
}
