// starway_amd._core — Python bindings.
//
// Class/method surface mirrors the reference's nanobind module signature for
// signature (reference src/starway/_bindings.pyi is the contract), with one
// extension: buffers may be host arrays (buffer protocol) OR device tensors
// (__cuda_array_interface__, e.g. torch HIP tensors) — the reference was
// CPU-only (nb::device::cpu, main.hpp:155) and left GPU "planned".
#include "core.hpp"

#include <pybind11/functional.h>
#include <pybind11/stl.h>

namespace sw {

namespace gpu {
int device_of(const void* ptr);  // gpu_ptr.cpp helper
}

// Parse a message buffer into (ptr, nbytes, device). Host path uses the
// buffer protocol (zero-copy, C-contiguous); device path uses
// __cuda_array_interface__ (torch ROCm tensors export it).
static BufferRef parse_buffer(py::handle obj, bool writable) {
  BufferRef ref;
  if (py::hasattr(obj, "__cuda_array_interface__")) {
    py::dict cai = obj.attr("__cuda_array_interface__").cast<py::dict>();
    py::tuple data = cai["data"].cast<py::tuple>();
    uintptr_t ptr = data[0].cast<uintptr_t>();
    bool readonly = data[1].cast<bool>();
    if (writable && readonly)
      throw std::invalid_argument("recv buffer is read-only");
    py::tuple shape = cai["shape"].cast<py::tuple>();
    std::string typestr = cai["typestr"].cast<std::string>();
    size_t itemsize = std::stoul(typestr.substr(2));
    size_t n = 1;
    for (auto d : shape) n *= d.cast<size_t>();
    uint64_t rows = 0, row_bytes = 0, row_stride = 0;
    if (cai.contains("strides") && !cai["strides"].is_none()) {
      py::tuple strides = cai["strides"].cast<py::tuple>();
      size_t expect = itemsize;
      bool contiguous = true;
      for (ssize_t i = (ssize_t)shape.size() - 1; i >= 0; --i) {
        if (strides[i].cast<size_t>() != expect) contiguous = false;
        expect *= shape[i].cast<size_t>();
      }
      if (!contiguous) {
        // 2D row-strided layout is supported natively (pack/unpack in the
        // pull kernel): last dim dense, leading dim strided.
        if (shape.size() == 2 &&
            strides[1].cast<size_t>() == itemsize) {
          rows = shape[0].cast<uint64_t>();
          row_bytes = shape[1].cast<uint64_t>() * itemsize;
          row_stride = strides[0].cast<uint64_t>();
          if (row_stride < row_bytes)
            throw std::invalid_argument(
                "overlapping strided device buffer is not supported");
        } else {
          throw std::invalid_argument(
              "device buffer must be contiguous or 2D row-strided "
              "(last dim dense) for zero-copy messaging");
        }
      }
    }
    ref.ptr = (uint8_t*)ptr;
    ref.size = n * itemsize;
    if (rows > 1) {
      ref.rows = rows;
      ref.row_bytes = row_bytes;
      ref.stride = row_stride;
    }
    ref.device = gpu::device_of((const void*)ptr);
    if (ref.device < 0)
      throw std::invalid_argument(
          "buffer exports __cuda_array_interface__ but is not resident on a "
          "visible HIP device");
    return ref;
  }
  Py_buffer view;
  int flags = PyBUF_C_CONTIGUOUS | (writable ? PyBUF_WRITABLE : PyBUF_SIMPLE);
  if (PyObject_GetBuffer(obj.ptr(), &view, flags) != 0)
    throw py::error_already_set();
  ref.ptr = (uint8_t*)view.buf;
  ref.size = (size_t)view.len;
  ref.device = -1;
  PyBuffer_Release(&view);  // the keepalive object pins the memory
  return ref;
}

struct PyServer {
  Engine engine{Engine::ServerMode};
  explicit PyServer(Context&) {
    engine.preferred_device_ = gpu::current_device();
  }
};

struct PyClient {
  Engine engine{Engine::ClientMode};
  explicit PyClient(Context&) {
    engine.preferred_device_ = gpu::current_device();
  }
};

}  // namespace sw

using namespace sw;

PYBIND11_MODULE(_core, m) {
  m.doc() =
      "starway_amd native core: MI355X-native tagged async zero-copy "
      "messaging (TCP control plane + hipIpc/xGMI data plane with gfx950 "
      "copy kernels)";

  py::class_<Context>(m, "Context").def(py::init<>());

  py::class_<EndpointInfo, std::shared_ptr<EndpointInfo>>(m, "ServerEndpoint")
      .def_property_readonly("name",
                             [](const EndpointInfo& e) { return e.name; })
      .def_property_readonly(
          "local_addr", [](const EndpointInfo& e) { return e.local_addr; })
      .def_property_readonly(
          "local_port", [](const EndpointInfo& e) { return e.local_port; })
      .def_property_readonly(
          "remote_addr", [](const EndpointInfo& e) { return e.remote_addr; })
      .def_property_readonly(
          "remote_port", [](const EndpointInfo& e) { return e.remote_port; })
      .def("view_transports",
           [](const EndpointInfo& e) { return e.transports; })
      .def("__eq__",
           [](const EndpointInfo& a, py::object other) {
             if (!py::isinstance<EndpointInfo>(other)) return false;
             return &a == other.cast<EndpointInfo*>();
           })
      .def("__hash__",
           [](const EndpointInfo& e) { return (size_t)(uintptr_t)&e; })
      .def("__repr__", [](const EndpointInfo& e) {
        return "<ServerEndpoint " + e.name + ">";
      });

  py::class_<PyServer>(m, "Server")
      .def(py::init<Context&>(), py::arg("ctx"), py::keep_alive<1, 2>())
      .def("set_accept_callback",
           [](PyServer& s, py::object cb) {
             s.engine.set_accept_callback(std::move(cb));
           },
           py::arg("callback"))
      .def("listen",
           [](PyServer& s, const std::string& addr, int port) {
             s.engine.listen(addr, port);
           },
           py::arg("addr"), py::arg("port"))
      .def("listen_address", [](PyServer& s) { s.engine.listen_address(); })
      .def("get_worker_address",
           [](PyServer& s) {
             auto v = s.engine.get_worker_address();
             return py::bytes((const char*)v.data(), v.size());
           })
      .def("close",
           [](PyServer& s, py::object cb) { s.engine.close(std::move(cb)); },
           py::arg("callback"))
      .def("send",
           [](PyServer& s, std::shared_ptr<EndpointInfo> ep, py::object buffer,
              uint64_t tag, py::object done, py::object fail) {
             BufferRef ref = parse_buffer(buffer, /*writable=*/false);
             s.engine.send(std::move(ep), ref, tag, std::move(done),
                           std::move(fail), std::move(buffer));
           },
           py::arg("client_ep"), py::arg("buffer"), py::arg("tag"),
           py::arg("done_callback"), py::arg("fail_callback"))
      .def("recv",
           [](PyServer& s, py::object buffer, uint64_t tag, uint64_t tag_mask,
              py::object done, py::object fail) {
             BufferRef ref = parse_buffer(buffer, /*writable=*/true);
             s.engine.recv(ref, tag, tag_mask, std::move(done),
                           std::move(fail), std::move(buffer));
           },
           py::arg("buffer"), py::arg("tag"), py::arg("tag_mask"),
           py::arg("done_callback"), py::arg("fail_callback"))
      .def("flush",
           [](PyServer& s, py::object done, py::object fail) {
             s.engine.flush(std::move(done), std::move(fail));
           },
           py::arg("done_callback"), py::arg("fail_callback"))
      .def("flush_ep",
           [](PyServer& s, std::shared_ptr<EndpointInfo> ep, py::object done,
              py::object fail) {
             s.engine.flush_ep(std::move(ep), std::move(done),
                               std::move(fail));
           },
           py::arg("client_ep"), py::arg("done_callback"),
           py::arg("fail_callback"))
      .def("get_stats",
           [](PyServer& s) {
             py::dict d;
             auto& st = s.engine.stats_;
             d["msgs_sent"] = st.msgs_sent.load();
             d["msgs_received"] = st.msgs_received.load();
             d["bytes_sent"] = st.bytes_sent.load();
             d["bytes_received"] = st.bytes_received.load();
             d["eager_rx"] = st.eager_rx.load();
             d["gpu_rx"] = st.gpu_rx.load();
             d["cma_rx"] = st.cma_rx.load();
             d["unexpected_rx"] = st.unexpected_rx.load();
             d["unexp_staged_bytes"] = st.unexp_staged_bytes.load();
             d["deferred_sends"] = st.deferred_sends.load();
             d["inbox_rx"] = st.inbox_rx.load();
             d["inbox_tx"] = st.inbox_tx.load();
             d["doorbell_rx"] = st.doorbell_rx.load();
             return d;
           })
      .def("list_clients",
           [](PyServer& s) {
             py::set out;
             for (auto& ep : s.engine.list_clients()) out.add(py::cast(ep));
             return out;
           })
      .def("evaluate_perf",
           [](PyServer& s, std::shared_ptr<EndpointInfo> ep,
              uint64_t msg_size) {
             return s.engine.evaluate_perf(std::move(ep), msg_size);
           },
           py::arg("client_ep"), py::arg("msg_size"));

  py::class_<PyClient>(m, "Client")
      .def(py::init<Context&>(), py::arg("ctx"), py::keep_alive<1, 2>())
      .def("connect",
           [](PyClient& c, const std::string& addr, int port, py::object cb) {
             c.engine.connect(addr, port, std::move(cb));
           },
           py::arg("addr"), py::arg("port"), py::arg("callback"))
      .def("connect_address",
           [](PyClient& c, py::bytes blob, py::object cb) {
             std::string s = blob;
             std::vector<uint8_t> v(s.begin(), s.end());
             c.engine.connect_address(v, std::move(cb));
           },
           py::arg("remote_address"), py::arg("callback"))
      .def("get_worker_address",
           [](PyClient& c) {
             auto v = c.engine.get_worker_address();
             return py::bytes((const char*)v.data(), v.size());
           })
      .def("close",
           [](PyClient& c, py::object cb) { c.engine.close(std::move(cb)); },
           py::arg("callback"))
      .def("send",
           [](PyClient& c, py::object buffer, uint64_t tag, py::object done,
              py::object fail) {
             BufferRef ref = parse_buffer(buffer, /*writable=*/false);
             c.engine.send(nullptr, ref, tag, std::move(done),
                           std::move(fail), std::move(buffer));
           },
           py::arg("buffer"), py::arg("tag"), py::arg("done_callback"),
           py::arg("fail_callback"))
      .def("recv",
           [](PyClient& c, py::object buffer, uint64_t tag, uint64_t tag_mask,
              py::object done, py::object fail) {
             BufferRef ref = parse_buffer(buffer, /*writable=*/true);
             c.engine.recv(ref, tag, tag_mask, std::move(done),
                           std::move(fail), std::move(buffer));
           },
           py::arg("buffer"), py::arg("tag"), py::arg("tag_mask"),
           py::arg("done_callback"), py::arg("fail_callback"))
      .def("flush",
           [](PyClient& c, py::object done, py::object fail) {
             c.engine.flush(std::move(done), std::move(fail));
           },
           py::arg("done_callback"), py::arg("fail_callback"))
      .def("get_stats",
           [](PyClient& c) {
             py::dict d;
             auto& st = c.engine.stats_;
             d["msgs_sent"] = st.msgs_sent.load();
             d["msgs_received"] = st.msgs_received.load();
             d["bytes_sent"] = st.bytes_sent.load();
             d["bytes_received"] = st.bytes_received.load();
             d["eager_rx"] = st.eager_rx.load();
             d["gpu_rx"] = st.gpu_rx.load();
             d["cma_rx"] = st.cma_rx.load();
             d["unexpected_rx"] = st.unexpected_rx.load();
             d["unexp_staged_bytes"] = st.unexp_staged_bytes.load();
             d["deferred_sends"] = st.deferred_sends.load();
             d["inbox_rx"] = st.inbox_rx.load();
             d["inbox_tx"] = st.inbox_tx.load();
             d["doorbell_rx"] = st.doorbell_rx.load();
             return d;
           })
      .def("evaluate_perf", [](PyClient& c, uint64_t msg_size) {
        return c.engine.evaluate_perf(nullptr, msg_size);
      },
           py::arg("msg_size"));

  m.def("gpu_available", [] { return gpu::available(); });
  // Run (or re-run) the on-box copy microbenchmark backing evaluate_perf
  // and persist it to ~/.cache/starway/perf.cal (STARWAY_CALIB_FILE to
  // override). Returns {same_gpu_gbps, xgmi_gbps} in effect.
  m.def("calibrate",
        [](bool force) {
          std::string err;
          bool ok;
          {
            py::gil_scoped_release rel;
            ok = gpu::calibrate(force, &err);
          }
          if (!ok) throw std::runtime_error("calibrate: " + err);
          py::dict d;
          d["same_gpu_gbps"] = gpu::same_gpu_copy_gbps();
          d["xgmi_gbps"] = gpu::xgmi_link_gbps();
          return d;
        },
        py::arg("force") = false);
  m.def("gpu_device_count", [] { return gpu::device_count(); });
  // IPC hygiene: close every imported hipIpc mapping and drop the handle
  // caches. Call after returning GPU memory to the driver (e.g.
  // torch.cuda.empty_cache()) so a reused base address can never serve a
  // stale handle. Also registered as an atexit hook by the Python facade.
  m.def("ipc_invalidate", [] {
    py::gil_scoped_release rel;
    gpu::ipc_close_all();
  });
  // Test/bench hook: run the gfx950 copy kernel dst<-src synchronously.
  m.def("_copy_device_sync",
        [](uintptr_t dst, uintptr_t src, size_t nbytes, int device) {
          std::string err;
          bool ok;
          {
            py::gil_scoped_release rel;
            ok = gpu::copy_device_sync((void*)dst, (const void*)src, nbytes,
                                       device, &err);
          }
          if (!ok) throw std::runtime_error("copy_device_sync: " + err);
        },
        py::arg("dst"), py::arg("src"), py::arg("nbytes"), py::arg("device"));
  m.attr("__version__") = "0.1.0";
}
