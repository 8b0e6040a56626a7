// starway_amd._rccl — RCCL-backed multi-endpoint fan-out.
//
// The engine's tagged path (engine.cpp) delivers each message with one
// hipIpc pull over a single xGMI link. For all-pairs / collective-shaped
// patterns RCCL's ncclSend/ncclRecv groups schedule traffic across all 7
// xGMI links per MI355X and route around busy links, so a mesh of
// endpoints can saturate the full ~1 TB/s per-GPU aggregate. This module
// exposes that as a thin group object; rendezvous of the ncclUniqueId is
// the caller's job (e.g. over starway tagged messages or torchrun/gloo).
//
// Kept separate from _core so importing starway_amd does not load librccl.
#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>
#include <rccl/rccl.h>

#include <stdexcept>
#include <string>

namespace py = pybind11;

namespace {

void check_nccl(ncclResult_t r, const char* what) {
  if (r != ncclSuccess)
    throw std::runtime_error(std::string(what) + ": " +
                             ncclGetErrorString(r));
}

void check_hip(hipError_t e, const char* what) {
  if (e != hipSuccess)
    throw std::runtime_error(std::string(what) + ": " + hipGetErrorString(e));
}

class RcclGroup {
 public:
  RcclGroup(py::bytes id_bytes, int rank, int world, int device)
      : rank_(rank), world_(world), device_(device) {
    std::string id = id_bytes;
    if (id.size() != sizeof(ncclUniqueId))
      throw std::invalid_argument("unique id must be " +
                                  std::to_string(sizeof(ncclUniqueId)) +
                                  " bytes");
    ncclUniqueId uid;
    memcpy(&uid, id.data(), sizeof(uid));
    check_hip(hipSetDevice(device), "hipSetDevice");
    check_hip(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking),
              "hipStreamCreate");
    {
      py::gil_scoped_release rel;  // collective init blocks on peers
      check_nccl(ncclCommInitRank(&comm_, world, uid, rank),
                 "ncclCommInitRank");
    }
  }

  ~RcclGroup() {
    if (comm_) ncclCommDestroy(comm_);
    if (stream_) hipStreamDestroy(stream_);
  }

  void group_start() { check_nccl(ncclGroupStart(), "ncclGroupStart"); }
  void group_end() {
    py::gil_scoped_release rel;
    check_nccl(ncclGroupEnd(), "ncclGroupEnd");
  }

  void send(uintptr_t ptr, size_t nbytes, int peer) {
    check_nccl(ncclSend((const void*)ptr, nbytes, ncclChar, peer, comm_,
                        stream_),
               "ncclSend");
  }

  void recv(uintptr_t ptr, size_t nbytes, int peer) {
    check_nccl(
        ncclRecv((void*)ptr, nbytes, ncclChar, peer, comm_, stream_),
        "ncclRecv");
  }

  // Equal-chunk all-to-all: send nbytes to every peer from sendbuf
  // (peer-major layout) and receive likewise into recvbuf.
  void all_to_all(uintptr_t sendbuf, uintptr_t recvbuf, size_t nbytes) {
    check_nccl(ncclGroupStart(), "ncclGroupStart");
    for (int p = 0; p < world_; p++) {
      check_nccl(ncclSend((const void*)(sendbuf + (size_t)p * nbytes),
                          nbytes, ncclChar, p, comm_, stream_),
                 "ncclSend");
      check_nccl(ncclRecv((void*)(recvbuf + (size_t)p * nbytes), nbytes,
                          ncclChar, p, comm_, stream_),
                 "ncclRecv");
    }
    check_nccl(ncclGroupEnd(), "ncclGroupEnd");
  }

  void synchronize() {
    py::gil_scoped_release rel;
    check_hip(hipStreamSynchronize(stream_), "hipStreamSynchronize");
  }

  int rank() const { return rank_; }
  int world() const { return world_; }

 private:
  ncclComm_t comm_ = nullptr;
  hipStream_t stream_ = nullptr;
  int rank_, world_, device_;
};

}  // namespace

PYBIND11_MODULE(_rccl, m) {
  m.doc() = "RCCL ncclSend/ncclRecv fan-out over the xGMI mesh";
  m.def("unique_id", [] {
    ncclUniqueId uid;
    check_nccl(ncclGetUniqueId(&uid), "ncclGetUniqueId");
    return py::bytes((const char*)&uid, sizeof(uid));
  });
  py::class_<RcclGroup>(m, "RcclGroup")
      .def(py::init<py::bytes, int, int, int>(), py::arg("unique_id"),
           py::arg("rank"), py::arg("world"), py::arg("device"))
      .def("group_start", &RcclGroup::group_start)
      .def("group_end", &RcclGroup::group_end)
      .def("send", &RcclGroup::send, py::arg("ptr"), py::arg("nbytes"),
           py::arg("peer"))
      .def("recv", &RcclGroup::recv, py::arg("ptr"), py::arg("nbytes"),
           py::arg("peer"))
      .def("all_to_all", &RcclGroup::all_to_all, py::arg("sendbuf"),
           py::arg("recvbuf"), py::arg("nbytes"))
      .def("synchronize", &RcclGroup::synchronize)
      .def_property_readonly("rank", &RcclGroup::rank)
      .def_property_readonly("world", &RcclGroup::world);
}
