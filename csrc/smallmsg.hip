// Small-message inbox data plane — gfx950 push/unpack/doorbell kernels.
//
// Replaces the RTS -> remote-pull -> RECV_DONE rendezvous for messages up
// to ~4 KiB with a sender-PUSH design sized for the 64 B latency and
// 64x1KiB throughput configs (BASELINE configs 4 and the small-messages
// scenario):
//
//   * each side owns a device "inbox" ring (allocated at HELLO, exported
//     once via hipIpc); the sender's push kernel writes payload + header
//     straight into the peer's ring over xGMI and publishes a per-slot
//     monotonic sequence word (system-scope release) — no host staging,
//     no per-message IPC traffic, no RECV_DONE ack
//   * delivery on the receiver is a LOCAL copy: either the engine's
//     batched unpack kernel (one launch + pinned-host result words for up
//     to 32 messages — the launch/event cost is what capped the round-1
//     rate at 85k msgs/s), or, on the latency path, a pre-armed doorbell
//     kernel that is already resident when the payload lands: it spins on
//     the next sequence word (relaxed system load + s_sleep, bounded),
//     tag-checks the header, copies into the posted recv buffer and
//     signals a pinned-host flag the progress thread polls — the receiver
//     side then costs ~1 kernel-resident copy instead of launch+event
//   * every spin is iteration-bounded (order 1-6 ms of device time) and
//     host-cancelable via a pinned cancel word, so no kernel can wedge the
//     GPU (re-arming is the engine's job)
//
// Memory-visibility discipline (cdna_hip_programming.md §6 G16, adapted to
// cross-device xGMI): payload with plain wide stores, every storing wave
// drains vmcnt, one lane publishes the sequence word with a system-scope
// release store; consumers poll relaxed and issue ONE system-scope acquire
// after the match, then use plain loads. Host-visible completion flags are
// system-scope release stores into hipHostMalloc memory.
//
// (Reference parity note: the reference had no GPU path at all — this
// subsystem replaces what UCX's eager-short protocol + cuda_copy transport
// would have done, reference benchmark.md:63-89.)
#include "core.hpp"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <map>
#include <mutex>
#include <tuple>
#include <vector>

namespace sw {
namespace gpu {

void* import_ipc(const uint8_t* handle, int open_device, std::string* err);

static std::mutex sm_mu;

// ---------------------------------------------------------------------------
// kernels
// ---------------------------------------------------------------------------

// Slot layout: [u64 seq][u64 tag][u64 size][40B pad][payload ...]
// seq is written LAST (release); its value IS the epoch, so slot reuse is
// ABA-safe without any reset traffic.

constexpr int kPushMax = 32;

struct PushDesc {
  const uint8_t* src;
  uint8_t* slot;
  uint32_t size;
  uint64_t seq;
  uint64_t tag;
};
struct PushArgs {
  PushDesc d[kPushMax];
  int n;
};

__global__ __launch_bounds__(256) void k_inbox_push(PushArgs args) {
  const PushDesc m = args.d[blockIdx.x];
  uint8_t* payload = m.slot + kInboxHdrBytes;
  if (threadIdx.x == 0) {
    *(unsigned long long*)(m.slot + 8) = m.tag;
    *(unsigned long long*)(m.slot + 16) = (unsigned long long)m.size;
  }
  uintptr_t sp = (uintptr_t)m.src;
  if ((sp & 15) == 0 && (m.size & 15) == 0) {
    const uint4* s4 = (const uint4*)sp;
    uint4* d4 = (uint4*)payload;
    for (uint32_t i = threadIdx.x; i < m.size / 16; i += blockDim.x)
      d4[i] = s4[i];
  } else {
    for (uint32_t i = threadIdx.x; i < m.size; i += blockDim.x)
      payload[i] = m.src[i];
  }
  // Publish: every storing wave drains, one lane releases the seq word.
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0)
    __hip_atomic_store((unsigned long long*)m.slot, m.seq, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_SYSTEM);
}

struct UnpackDesc {
  const uint8_t* slot;
  uint8_t* dst;
  uint32_t size;
  uint64_t seq;
};
struct UnpackArgs {
  UnpackDesc d[kPushMax];
  unsigned long long* results;  // pinned host, one per message
  int n;
  unsigned spin_iters;
};

// result word: 0 pending / 1 ok / 2 payload-never-arrived
__global__ __launch_bounds__(256) void k_inbox_unpack(UnpackArgs args) {
  const UnpackDesc m = args.d[blockIdx.x];
  __shared__ int ok;
  if (threadIdx.x == 0) {
    ok = 0;
    for (unsigned it = 0; it < args.spin_iters; ++it) {
      unsigned long long s = __hip_atomic_load(
          (const unsigned long long*)m.slot, __ATOMIC_RELAXED,
          __HIP_MEMORY_SCOPE_SYSTEM);
      if (s == m.seq) {
        ok = 1;
        break;
      }
      __builtin_amdgcn_s_sleep(8);
    }
    if (ok)  // ONE acquire after the match drops stale lines
      (void)__hip_atomic_load((const unsigned long long*)m.slot,
                              __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
  }
  __syncthreads();
  if (ok) {
    const uint8_t* payload = m.slot + kInboxHdrBytes;
    uintptr_t dp = (uintptr_t)m.dst;
    if ((dp & 15) == 0 && (m.size & 15) == 0) {
      const uint4* s4 = (const uint4*)payload;
      uint4* d4 = (uint4*)dp;
      for (uint32_t i = threadIdx.x; i < m.size / 16; i += blockDim.x)
        d4[i] = s4[i];
    } else {
      for (uint32_t i = threadIdx.x; i < m.size; i += blockDim.x)
        m.dst[i] = payload[i];
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __syncthreads();
  if (threadIdx.x == 0)
    // Release store: flushes the dst writes before the host sees the flag.
    __hip_atomic_store(&args.results[blockIdx.x], ok ? 1ull : 2ull,
                       __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}

struct ArmArgs {
  const uint8_t* slot;  // inbox slot of the expected sequence number
  uint64_t expect_seq;
  uint64_t tag;
  uint64_t mask;
  uint8_t* dst;
  uint64_t max_size;
  unsigned long long* result;  // pinned host
  unsigned int* cancel;        // pinned host
  unsigned spin_iters;
};

// result word: 0 running / 1|size<<32 copied / 2 nomatch / 3 canceled /
// 4 expired. The kernel only CONSUMES a message whose header matches its
// armed (tag, mask) and fits the buffer; anything else is left in the slot
// for the engine's ordinary matching path.
__global__ __launch_bounds__(256) void k_inbox_wait(ArmArgs a) {
  __shared__ unsigned long long st;
  if (threadIdx.x == 0) {
    unsigned long long status = 4;  // expired unless something happens
    for (unsigned it = 0; it < a.spin_iters; ++it) {
      unsigned long long s = __hip_atomic_load(
          (const unsigned long long*)a.slot, __ATOMIC_RELAXED,
          __HIP_MEMORY_SCOPE_SYSTEM);
      if (s == a.expect_seq) {
        (void)__hip_atomic_load((const unsigned long long*)a.slot,
                                __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
        unsigned long long mtag = *(const unsigned long long*)(a.slot + 8);
        unsigned long long msz = *(const unsigned long long*)(a.slot + 16);
        status = ((mtag & a.mask) == (a.tag & a.mask) && msz <= a.max_size)
                     ? (1ull | (msz << 32))
                     : 2ull;
        break;
      }
      if ((it & 63) == 63) {  // PCIe poll of the cancel word, amortized
        if (__hip_atomic_load(a.cancel, __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_SYSTEM)) {
          status = 3;
          break;
        }
      }
      __builtin_amdgcn_s_sleep(8);
    }
    st = status;
  }
  __syncthreads();
  unsigned long long status = st;
  if ((status & 0xFF) == 1) {
    uint32_t size = (uint32_t)(status >> 32);
    const uint8_t* payload = a.slot + kInboxHdrBytes;
    uintptr_t dp = (uintptr_t)a.dst;
    if ((dp & 15) == 0 && (size & 15) == 0) {
      const uint4* s4 = (const uint4*)payload;
      uint4* d4 = (uint4*)dp;
      for (uint32_t i = threadIdx.x; i < size / 16; i += blockDim.x)
        d4[i] = s4[i];
    } else {
      for (uint32_t i = threadIdx.x; i < size; i += blockDim.x)
        a.dst[i] = payload[i];
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __syncthreads();
  if (threadIdx.x == 0)
    __hip_atomic_store(a.result, status, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_SYSTEM);
}

// ---------------------------------------------------------------------------
// host side: streams, pinned pool, events
// ---------------------------------------------------------------------------

enum class SmStream { Push, Unpack, Wait };

// Streams are keyed by (device, kind, lane). The lane is the calling
// engine's identity (mod 8): two engines in one process (loopback pairs)
// must NEVER share a Wait stream — a doorbell kernel spins for up to its
// bound, and a second engine's doorbell queued behind it on the same
// stream cannot start, serializing bidirectional latency paths into the
// expiry cadence (measured: pingpong-flag RTT 116 us median, 5 ms tail).
// STARWAY_SM_LANES caps the per-kind stream fan-out (power of two, max
// 8). Shared-GPU emulation (many engines in few processes) sets 1 so the
// stream count stays under the hardware-queue budget; the production
// 1-process-per-GPU topology keeps distinct per-engine lanes.
static int sm_lanes() {
  static const int n = [] {
    int v = (int)env_u64("STARWAY_SM_LANES", 8);
    int p2 = 1;
    while (p2 < v && p2 < 8) p2 <<= 1;
    return p2;
  }();
  return n;
}

static hipStream_t sm_stream(int device, SmStream kind, int lane) {
  static std::map<std::tuple<int, int, int>, hipStream_t> streams;  // sm_mu
  auto key = std::make_tuple(device, (int)kind, lane & (sm_lanes() - 1));
  auto it = streams.find(key);
  if (it != streams.end()) return it->second;
  int prev;
  hipGetDevice(&prev);
  hipSetDevice(device);
  hipStream_t s;
  if (hipStreamCreateWithFlags(&s, hipStreamNonBlocking) != hipSuccess)
    s = nullptr;
  hipSetDevice(prev);
  streams[key] = s;
  return s;
}

// Pinned-host cell pool: 64 B cells for flags/results, 4 KiB cells for
// host-recv bounces. hipHostMalloc costs tens of microseconds; the pool
// makes per-message pinned state free.
struct PinPool {
  std::vector<void*> arenas;
  std::vector<uint8_t*> free_cells;
  size_t cell = 0;

  uint8_t* get() {
    if (free_cells.empty()) {
      void* arena = nullptr;
      size_t n = 256;
      if (hipHostMalloc(&arena, cell * n, hipHostMallocDefault) !=
          hipSuccess)
        return nullptr;
      memset(arena, 0, cell * n);
      arenas.push_back(arena);
      for (size_t i = 0; i < n; i++)
        free_cells.push_back((uint8_t*)arena + i * cell);
    }
    uint8_t* p = free_cells.back();
    free_cells.pop_back();
    memset(p, 0, cell);
    return p;
  }
  void put(uint8_t* p) { free_cells.push_back(p); }
};
static PinPool g_flag_pool{.cell = 64};
// Bounce cells sized to the inbox payload cap so a host-destined message
// always fits its staging cell (also holds the 512-entry results block).
static PinPool g_bounce_pool{.cell = 0};
static void init_pools() {
  if (!g_bounce_pool.cell)
    g_bounce_pool.cell =
        std::max<size_t>(4096, (env_u64("STARWAY_INBOX_MAX", 4096) + 63) &
                                   ~63ull);
}

static std::vector<hipEvent_t> g_sm_events;  // sm_mu held

static hipError_t sm_event(int device, hipEvent_t* ev) {
  if (!g_sm_events.empty()) {
    *ev = g_sm_events.back();
    g_sm_events.pop_back();
    return hipSuccess;
  }
  (void)device;
  return hipEventCreateWithFlags(ev, hipEventDisableTiming);
}

// ---------------------------------------------------------------------------
// inbox lifecycle
// ---------------------------------------------------------------------------

bool inbox_create(InboxInfo* out, int device, std::string* err) {
  std::lock_guard<std::mutex> lk(sm_mu);
  // The engine's progress thread never calls hipSetDevice, so its current
  // device is always 0 — the ring must land on the device the USER works
  // on (captured at Server/Client construction on the Python thread; one
  // process per GPU in the bench/production topology).
  int prev = 0;
  hipGetDevice(&prev);
  int dev = device >= 0 ? device : prev;
  if (hipSetDevice(dev) != hipSuccess) {
    *err = "no HIP device";
    return false;
  }
  static const uint32_t slots = (uint32_t)env_u64("STARWAY_INBOX_SLOTS", 64);
  static const uint32_t payload =
      (uint32_t)env_u64("STARWAY_INBOX_MAX", 4096);
  uint32_t slot_bytes = kInboxHdrBytes + ((payload + 63) & ~63u);
  void* base = nullptr;
  hipError_t e = hipMalloc(&base, (size_t)slots * slot_bytes);
  if (e != hipSuccess) {
    hipSetDevice(prev);
    *err = std::string("inbox alloc: ") + hipGetErrorString(e);
    return false;
  }
  hipMemset(base, 0, (size_t)slots * slot_bytes);
  hipDeviceSynchronize();  // seq words must read 0 before the export leaks
  hipIpcMemHandle_t h;
  e = hipIpcGetMemHandle(&h, base);
  hipSetDevice(prev);
  if (e != hipSuccess) {
    hipFree(base);
    *err = std::string("inbox export: ") + hipGetErrorString(e);
    return false;
  }
  out->base = (uint64_t)(uintptr_t)base;
  out->device = dev;
  out->slots = slots;
  out->slot_bytes = slot_bytes;
  memcpy(out->handle, &h, kIpcHandleBytes);
  // Pre-warm this translation unit's code object (lazy hipModule load
  // costs ~10-17 ms on first launch — pay it here at connection setup,
  // not on the first message). Benign launches: seq 0 is the empty
  // sentinel, sizes are 0, and the scratch flag cell is pool-local.
  init_pools();
  if (uint8_t* cell = g_flag_pool.get()) {
    hipStream_t s = sm_stream(dev, SmStream::Push, 0);
    PushArgs pa{};
    pa.n = 1;
    pa.d[0] = PushDesc{(const uint8_t*)base, (uint8_t*)base, 0, 0, 0};
    hipLaunchKernelGGL(k_inbox_push, dim3(1), dim3(256), 0, s, pa);
    UnpackArgs ua{};
    ua.n = 1;
    ua.results = (unsigned long long*)cell;
    ua.spin_iters = 1;
    ua.d[0] = UnpackDesc{(const uint8_t*)base, (uint8_t*)base, 0, 0};
    hipLaunchKernelGGL(k_inbox_unpack, dim3(1), dim3(256), 0, s, ua);
    ArmArgs aa{};
    aa.slot = (const uint8_t*)base;
    aa.expect_seq = ~0ull;  // never matches: expires after one spin
    aa.dst = (uint8_t*)base;
    aa.result = (unsigned long long*)cell;
    aa.cancel = (unsigned int*)(cell + 8);
    aa.spin_iters = 1;
    hipLaunchKernelGGL(k_inbox_wait, dim3(1), dim3(256), 0, s, aa);
    hipStreamSynchronize(s);
    (void)hipGetLastError();
    g_flag_pool.put(cell);
  }
  return true;
}

void inbox_destroy(const InboxInfo& ib) {
  std::lock_guard<std::mutex> lk(sm_mu);
  int prev;
  hipGetDevice(&prev);
  hipSetDevice(ib.device);
  hipFree((void*)(uintptr_t)ib.base);
  hipSetDevice(prev);
}

// Resolve the peer's inbox base in OUR address space: raw pointer when the
// peer is this very process (loopback), else the cached hipIpc mapping.
static uint8_t* peer_inbox_base(const InboxInfo& peer, bool same_proc,
                                int run_device, std::string* err) {
  if (same_proc) return (uint8_t*)(uintptr_t)peer.base;
  return (uint8_t*)import_ipc(peer.handle, run_device, err);
}

// ---------------------------------------------------------------------------
// push
// ---------------------------------------------------------------------------

struct PushTicket {
  hipEvent_t ev = nullptr;
  int device = -1;
};

void* inbox_push(const InboxInfo& peer, bool same_proc, int run_device,
                 const PushMsg* msgs, int n, int lane, std::string* err) {
  std::lock_guard<std::mutex> lk(sm_mu);
  if (n < 1 || n > kPushMax) {
    *err = "inbox_push: bad batch size";
    return nullptr;
  }
  uint8_t* base = peer_inbox_base(peer, same_proc, run_device, err);
  if (!base) return nullptr;
  PushArgs args;
  args.n = n;
  for (int i = 0; i < n; i++) {
    args.d[i].src = msgs[i].src;
    args.d[i].slot = base + (msgs[i].seq % peer.slots) * peer.slot_bytes;
    args.d[i].size = msgs[i].size;
    args.d[i].seq = msgs[i].seq;
    args.d[i].tag = msgs[i].tag;
  }
  int prev;
  hipGetDevice(&prev);
  hipSetDevice(run_device);
  hipStream_t stream = sm_stream(run_device, SmStream::Push, lane);
  hipLaunchKernelGGL(k_inbox_push, dim3(n), dim3(256), 0, stream, args);
  hipError_t e = hipGetLastError();
  PushTicket* t = nullptr;
  if (e == hipSuccess) {
    t = new PushTicket();
    t->device = run_device;
    e = sm_event(run_device, &t->ev);
    if (e == hipSuccess) e = hipEventRecord(t->ev, stream);
  }
  hipSetDevice(prev);
  if (e != hipSuccess) {
    *err = std::string("inbox push: ") + hipGetErrorString(e);
    delete t;
    return nullptr;
  }
  return t;
}

int push_poll(void* ticket, std::string* err) {
  PushTicket* t = (PushTicket*)ticket;
  hipError_t e = hipEventQuery(t->ev);
  if (e == hipSuccess) return 1;
  if (e == hipErrorNotReady) return 0;
  *err = std::string("push failed: ") + hipGetErrorString(e);
  return -1;
}

void push_free(void* ticket) {
  PushTicket* t = (PushTicket*)ticket;
  if (t->ev) {
    std::lock_guard<std::mutex> lk(sm_mu);
    if (g_sm_events.size() < 256)
      g_sm_events.push_back(t->ev);
    else
      hipEventDestroy(t->ev);
  }
  delete t;
}

// ---------------------------------------------------------------------------
// unpack
// ---------------------------------------------------------------------------

struct UnpackTicket {
  unsigned long long* results = nullptr;  // one pinned 4 KiB block
  std::vector<uint8_t*> bounces;          // per message, null = direct dst
  int n = 0;
};

void* inbox_unpack(const InboxInfo& mine, const UnpackMsg* msgs, int n,
                   int lane, std::string* err) {
  std::lock_guard<std::mutex> lk(sm_mu);
  init_pools();
  if (n < 1 || n > kPushMax) {
    *err = "inbox_unpack: bad batch size";
    return nullptr;
  }
  auto* t = new UnpackTicket();
  t->n = n;
  t->results = (unsigned long long*)g_bounce_pool.get();  // 512 slots
  if (!t->results) {
    delete t;
    *err = "pinned pool exhausted";
    return nullptr;
  }
  static const unsigned spin =
      (unsigned)env_u64("STARWAY_UNPACK_SPIN", 30000);  // ~6 ms bound

  UnpackArgs args;
  args.n = n;
  args.results = t->results;
  args.spin_iters = spin;
  uint8_t* base = (uint8_t*)(uintptr_t)mine.base;
  for (int i = 0; i < n; i++) {
    args.d[i].slot = base + (msgs[i].seq % mine.slots) * mine.slot_bytes;
    args.d[i].seq = msgs[i].seq;
    args.d[i].size = msgs[i].size;
    uint8_t* dst = msgs[i].dst;
    if (!dst) {
      dst = g_bounce_pool.get();  // host recv: pinned staging
      if (!dst) {
        g_bounce_pool.put((uint8_t*)t->results);
        for (auto* b : t->bounces)
          if (b) g_bounce_pool.put(b);
        delete t;
        *err = "pinned pool exhausted";
        return nullptr;
      }
    }
    args.d[i].dst = dst;
    t->bounces.push_back(msgs[i].dst ? nullptr : dst);
  }
  int prev;
  hipGetDevice(&prev);
  hipSetDevice(mine.device);
  hipStream_t stream = sm_stream(mine.device, SmStream::Unpack, lane);
  hipLaunchKernelGGL(k_inbox_unpack, dim3(n), dim3(256), 0, stream, args);
  hipError_t e = hipGetLastError();
  hipSetDevice(prev);
  if (e != hipSuccess) {
    *err = std::string("inbox unpack: ") + hipGetErrorString(e);
    g_bounce_pool.put((uint8_t*)t->results);
    for (auto* b : t->bounces)
      if (b) g_bounce_pool.put(b);
    delete t;
    return nullptr;
  }
  return t;
}

int unpack_poll(void* ticket, int idx, std::string* err) {
  UnpackTicket* t = (UnpackTicket*)ticket;
  unsigned long long v =
      __atomic_load_n(t->results + idx, __ATOMIC_ACQUIRE);
  if (v == 0) return 0;
  if (v == 1) return 1;
  *err = "inbox payload never arrived (peer died mid-push?)";
  return -1;
}

const uint8_t* unpack_bounce(void* ticket, int idx) {
  return ((UnpackTicket*)ticket)->bounces[idx];
}

void unpack_free(void* ticket) {
  UnpackTicket* t = (UnpackTicket*)ticket;
  std::lock_guard<std::mutex> lk(sm_mu);
  g_bounce_pool.put((uint8_t*)t->results);
  for (auto* b : t->bounces)
    if (b) g_bounce_pool.put(b);
  delete t;
}

// ---------------------------------------------------------------------------
// doorbell (armed wait)
// ---------------------------------------------------------------------------

struct ArmTicket {
  unsigned long long* result = nullptr;  // pinned
  unsigned int* cancel = nullptr;        // pinned (same cell, offset 8)
  int device = -1;
};

void* arm_recv(const InboxInfo& mine, uint64_t expect_seq, uint64_t tag,
               uint64_t mask, uint8_t* dst, uint64_t max_size, int lane,
               std::string* err) {
  std::lock_guard<std::mutex> lk(sm_mu);
  init_pools();
  uint8_t* cell = g_flag_pool.get();
  if (!cell) {
    *err = "pinned pool exhausted";
    return nullptr;
  }
  auto* t = new ArmTicket();
  t->result = (unsigned long long*)cell;
  t->cancel = (unsigned int*)(cell + 8);
  t->device = mine.device;
  static const unsigned spin =
      (unsigned)env_u64("STARWAY_ARM_SPIN", 8000);  // ~1 ms bound
  ArmArgs a;
  a.slot = (const uint8_t*)(uintptr_t)mine.base +
           (expect_seq % mine.slots) * mine.slot_bytes;
  a.expect_seq = expect_seq;
  a.tag = tag;
  a.mask = mask;
  a.dst = dst;
  a.max_size = max_size;
  a.result = t->result;
  a.cancel = t->cancel;
  a.spin_iters = spin;
  int prev;
  hipGetDevice(&prev);
  hipSetDevice(mine.device);
  hipStream_t stream = sm_stream(mine.device, SmStream::Wait, lane);
  hipLaunchKernelGGL(k_inbox_wait, dim3(1), dim3(256), 0, stream, a);
  hipError_t e = hipGetLastError();
  hipSetDevice(prev);
  if (e != hipSuccess) {
    *err = std::string("arm: ") + hipGetErrorString(e);
    g_flag_pool.put(cell);
    delete t;
    return nullptr;
  }
  return t;
}

int arm_poll(void* ticket, uint64_t* size_out) {
  ArmTicket* t = (ArmTicket*)ticket;
  unsigned long long v = __atomic_load_n(t->result, __ATOMIC_ACQUIRE);
  int status = (int)(v & 0xFF);
  if (status == 1 && size_out) *size_out = v >> 32;
  return status;  // 0 running / 1 copied / 2 nomatch / 3 canceled / 4 expired
}

void arm_cancel(void* ticket) {
  ArmTicket* t = (ArmTicket*)ticket;
  __atomic_store_n(t->cancel, 1u, __ATOMIC_RELEASE);
}

void arm_free(void* ticket) {
  ArmTicket* t = (ArmTicket*)ticket;
  std::lock_guard<std::mutex> lk(sm_mu);
  g_flag_pool.put((uint8_t*)t->result);
  delete t;
}

// The wait kernel never started (launch-queue backlog) and may still write
// its result later: the pinned cell must NOT return to the pool. Leaks one
// 64 B cell — correctness over thrift.
void arm_leak(void* ticket) { delete (ArmTicket*)ticket; }

}  // namespace gpu
}  // namespace sw
