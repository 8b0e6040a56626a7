// Shared-memory byte channel for same-host connections — the "sm"
// transport analog of the reference's UCX transport selection
// (benchmark.md there discusses tcp/sm/rc). Two SPSC byte rings in one
// /dev/shm segment carry the exact same frame stream as the TCP path;
// the engine's hot poll loop consumes them, so a same-host frame hop
// costs two memcpys + an atomic instead of a socket round trip.
//
// Handover protocol (engine.cpp): server creates the segment after HELLO
// when peer host-id matches, sends SHM_OFFER(name) over TCP; client maps
// it and replies SHM_ACK as its LAST TCP frame (sender-side TxItems carry
// a per-item channel flag, so the switch point is exact); server then
// sends SHM_SWITCH as ITS last TCP frame. Each receiver flips its rx
// source when it parses the peer's marker — frame order is preserved
// across the switch because both channels are drained by one parser.
//
// Liveness: no doorbells — the engine spins hot for ~50k iterations
// before sleeping at most 1 ms, so a message landing in an idle ring is
// picked up within 1 ms (the TCP socket stays open for EOF detection and
// the close path marks the ring `closed`).
#pragma once

#include <atomic>
#include <cstdint>
#include <cstring>
#include <fcntl.h>
#include <string>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

namespace sw {

struct ShmRingHdr {
  std::atomic<uint64_t> head;  // producer position (monotonic)
  char _p1[56];
  std::atomic<uint64_t> tail;  // consumer position (monotonic)
  char _p2[56];
  std::atomic<uint32_t> closed;
  char _p3[60];
};
static_assert(sizeof(ShmRingHdr) == 192);

// One direction of the channel (a view into the mapped segment).
struct ShmRingView {
  ShmRingHdr* h = nullptr;
  uint8_t* data = nullptr;
  uint64_t cap = 0;

  uint64_t readable() const {
    return h->head.load(std::memory_order_acquire) -
           h->tail.load(std::memory_order_relaxed);
  }
  uint64_t writable() const {
    return cap - (h->head.load(std::memory_order_relaxed) -
                  h->tail.load(std::memory_order_acquire));
  }
  bool peer_closed() const {
    return h->closed.load(std::memory_order_acquire) != 0;
  }
  void mark_closed() { h->closed.store(1, std::memory_order_release); }

  // Nonblocking read of up to n bytes; returns bytes read.
  size_t read(void* dst, size_t n) {
    uint64_t avail = readable();
    if (!avail) return 0;
    if (n > avail) n = avail;
    uint64_t tail = h->tail.load(std::memory_order_relaxed);
    uint64_t off = tail & (cap - 1);
    size_t first = (size_t)std::min<uint64_t>(n, cap - off);
    memcpy(dst, data + off, first);
    if (n > first) memcpy((uint8_t*)dst + first, data, n - first);
    h->tail.store(tail + n, std::memory_order_release);
    return n;
  }

  // Nonblocking write of up to n bytes; returns bytes written.
  size_t write(const void* src, size_t n) {
    uint64_t space = writable();
    if (!space) return 0;
    if (n > space) n = space;
    uint64_t head = h->head.load(std::memory_order_relaxed);
    uint64_t off = head & (cap - 1);
    size_t first = (size_t)std::min<uint64_t>(n, cap - off);
    memcpy(data + off, src, first);
    if (n > first) memcpy(data, (const uint8_t*)src + first, n - first);
    h->head.store(head + n, std::memory_order_release);
    return n;
  }
};

// The full segment: [hdr A][data A][hdr B][data B].
// Creator (server side) transmits on ring A, receives on ring B.
class ShmChannel {
 public:
  static size_t segment_size(uint64_t cap) {
    return 2 * (sizeof(ShmRingHdr) + cap);
  }

  // Create + map a fresh segment (server side).
  static ShmChannel* create(const std::string& name, uint64_t cap,
                            std::string* err) {
    int fd = shm_open(name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
    if (fd < 0) {
      *err = "shm_open(create): " + std::string(strerror(errno));
      return nullptr;
    }
    size_t sz = segment_size(cap);
    if (ftruncate(fd, (off_t)sz) != 0) {
      *err = "ftruncate: " + std::string(strerror(errno));
      ::close(fd);
      shm_unlink(name.c_str());
      return nullptr;
    }
    void* base = mmap(nullptr, sz, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    ::close(fd);
    if (base == MAP_FAILED) {
      *err = "mmap: " + std::string(strerror(errno));
      shm_unlink(name.c_str());
      return nullptr;
    }
    memset(base, 0, 2 * sizeof(ShmRingHdr));
    auto* ch = new ShmChannel(name, base, sz, cap, /*creator=*/true);
    return ch;
  }

  // Map an existing segment (client side).
  static ShmChannel* open(const std::string& name, uint64_t cap,
                          std::string* err) {
    int fd = shm_open(name.c_str(), O_RDWR, 0600);
    if (fd < 0) {
      *err = "shm_open: " + std::string(strerror(errno));
      return nullptr;
    }
    size_t sz = segment_size(cap);
    struct stat st {};
    if (fstat(fd, &st) != 0 || (size_t)st.st_size < sz) {
      *err = "shm segment size mismatch";
      ::close(fd);
      return nullptr;
    }
    void* base = mmap(nullptr, sz, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    ::close(fd);
    if (base == MAP_FAILED) {
      *err = "mmap: " + std::string(strerror(errno));
      return nullptr;
    }
    return new ShmChannel(name, base, sz, cap, /*creator=*/false);
  }

  ~ShmChannel() {
    tx.mark_closed();
    munmap(base_, size_);
    if (creator_) shm_unlink(name_.c_str());
  }

  ShmRingView tx, rx;
  const std::string& name() const { return name_; }

 private:
  ShmChannel(std::string name, void* base, size_t size, uint64_t cap,
             bool creator)
      : name_(std::move(name)), base_(base), size_(size), creator_(creator) {
    auto* a_hdr = (ShmRingHdr*)base;
    uint8_t* a_data = (uint8_t*)base + sizeof(ShmRingHdr);
    auto* b_hdr = (ShmRingHdr*)(a_data + cap);
    uint8_t* b_data = (uint8_t*)(b_hdr + 1);
    ShmRingView a{a_hdr, a_data, cap}, b{b_hdr, b_data, cap};
    if (creator) {
      tx = a;
      rx = b;
    } else {
      tx = b;
      rx = a;
    }
  }

  std::string name_;
  void* base_;
  size_t size_;
  bool creator_;
};

}  // namespace sw
