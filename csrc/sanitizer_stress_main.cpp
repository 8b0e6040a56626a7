// Sanitizer stress binary (TSAN/ASAN) — drives the engine's cross-thread
// contracts under instrumentation (SURVEY §5 race-detection item: "add a
// TSAN preset since we own the transport now").
//
// Embeds a Python interpreter (the engine's completion plumbing holds the
// GIL at controlled points; the GIL's pthread primitives give TSAN the
// happens-before edges it needs), then hammers the host transport:
//   * loopback Server/Client over TCP + shm-ring handover
//   * several submitter threads posting sends/recvs/flushes concurrently
//     against the two engine progress threads
//   * connect/close churn with in-flight ops (teardown/cancel races)
// Exit code 0 = all expected completions arrived and no sanitizer report
// aborted the run (TSAN halt_on_error / default abort on ASAN).
//
// Build + run: python build_ext.py --tsan   (or scripts/sanitize.sh)
#include "core.hpp"

#include <pybind11/embed.h>

#include <atomic>
#include <chrono>
#include <cstdio>
#include <thread>
#include <vector>

using namespace sw;

static py::object make_cb(std::atomic<uint64_t>* counter) {
  return py::cpp_function([counter](py::args) {
    counter->fetch_add(1, std::memory_order_relaxed);
  });
}

int main() {
  py::scoped_interpreter guard;
  std::atomic<uint64_t> sends_done{0}, recvs_done{0}, fails{0}, flushes{0};

  {
    Engine server(Engine::ServerMode);
    Engine client(Engine::ClientMode);
    // Main holds the GIL after interpreter start: release it for the whole
    // driving phase (worker threads and the engines' completion plumbing
    // acquire it as needed); the release guard's destructor re-acquires
    // BEFORE the engines' destructors run (reverse declaration order).
    py::gil_scoped_release main_rel;
    {
      py::gil_scoped_acquire gil;
      server.listen("127.0.0.1", 0);
    }
    // Connect via the worker-address blob (exercises the candidate-route
    // path) — poll until running.
    std::vector<uint8_t> blob;
    {
      py::gil_scoped_acquire gil;
      blob = server.get_worker_address();
      client.connect_address(blob, make_cb(&flushes));
    }
    while (client.status() != 2) {
      if (client.status() == 4) {
        fprintf(stderr, "connect failed\n");
        return 1;
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }

    constexpr int kThreads = 4;
    constexpr int kMsgs = 1500;  // per thread
    std::vector<std::vector<uint8_t>> bufs(kThreads,
                                           std::vector<uint8_t>(512, 7));
    std::vector<std::vector<uint8_t>> rbufs(kThreads,
                                            std::vector<uint8_t>(512));

    // Receiver-poster thread: posts wildcard recvs on the server.
    std::thread recv_thread([&] {
      for (int i = 0; i < kThreads * kMsgs; i++) {
        py::gil_scoped_acquire gil;
        BufferRef ref;
        ref.ptr = rbufs[i % kThreads].data();
        ref.size = rbufs[i % kThreads].size();
        server.recv(ref, 0, 0, make_cb(&recvs_done), make_cb(&fails),
                    py::none());
      }
    });

    // Sender threads: tagged sends + periodic flushes from many threads
    // (the MPSC command-queue contract under contention).
    std::vector<std::thread> senders;
    for (int t = 0; t < kThreads; t++) {
      senders.emplace_back([&, t] {
        for (int i = 0; i < kMsgs; i++) {
          py::gil_scoped_acquire gil;
          BufferRef ref;
          ref.ptr = bufs[t].data();
          ref.size = bufs[t].size();
          client.send(nullptr, ref, (uint64_t)(t * kMsgs + i),
                      make_cb(&sends_done), make_cb(&fails), py::none());
          if (i % 256 == 255)
            client.flush(make_cb(&flushes), make_cb(&fails));
        }
      });
    }
    for (auto& th : senders) th.join();
    recv_thread.join();

    auto deadline =
        std::chrono::steady_clock::now() + std::chrono::seconds(60);
    uint64_t want = (uint64_t)kThreads * kMsgs;
    while ((sends_done.load() < want || recvs_done.load() < want) &&
           std::chrono::steady_clock::now() < deadline) {
      std::this_thread::sleep_for(std::chrono::milliseconds(5));
    }
    {
      py::gil_scoped_acquire gil;
      client.close(make_cb(&flushes));  // counters outlive the engines
      server.close(make_cb(&flushes));
    }
    // Engines join in their destructors (end of scope).
    if (sends_done.load() < want || recvs_done.load() < want) {
      fprintf(stderr, "stress incomplete: sends %llu recvs %llu of %llu\n",
              (unsigned long long)sends_done.load(),
              (unsigned long long)recvs_done.load(),
              (unsigned long long)want);
      return 2;
    }
  }

  // Churn phase: rapid connect/close with in-flight ops (teardown races).
  for (int round = 0; round < 8; round++) {
    std::atomic<uint64_t> dummy{0};  // outlives the engines' teardown cbs
    std::vector<uint8_t> blob;
    Engine server(Engine::ServerMode);
    Engine client(Engine::ClientMode);
    py::gil_scoped_release round_rel;
    {
      py::gil_scoped_acquire gil;
      server.listen("127.0.0.1", 0);
      blob = server.get_worker_address();
      client.connect_address(blob, make_cb(&dummy));
    }
    while (client.status() == 1)
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    if (client.status() == 2) {
      py::gil_scoped_acquire gil;
      static std::vector<uint8_t> payload(1 << 20, 9);
      BufferRef ref;
      ref.ptr = payload.data();
      ref.size = payload.size();
      client.send(nullptr, ref, 1, make_cb(&dummy), make_cb(&dummy),
                  py::none());
      client.close(make_cb(&dummy));  // close with the send in flight
      server.close(make_cb(&dummy));
    }
  }

  printf("sanitizer stress OK: %llu sends, %llu recvs, %llu flushes\n",
         (unsigned long long)sends_done.load(),
         (unsigned long long)recvs_done.load(),
         (unsigned long long)flushes.load());
  return 0;
}
