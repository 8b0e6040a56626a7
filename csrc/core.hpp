// starway_amd core — MI355X-native tagged async zero-copy messaging engine.
//
// Brand-new design with the capability surface of Clouder0/starway
// (reference: /root/reference/src/bindings/main.hpp — API contract only; the
// transport below is our own: TCP control/CPU-data plane + hipIpc/xGMI GPU
// data plane with hand-written gfx950 copy kernels, no UCX).
//
// Architecture invariants (mirrors reference main.cpp:150-161 design stance):
//  * all transport state is owned by ONE progress thread per Client/Server
//  * the Python thread communicates with it only through a command queue and
//    atomic status flags
//  * Python callbacks fire only from controlled points with the GIL held
#pragma once

#include <arpa/inet.h>
#include <atomic>
#include <cstdint>
#include <cstring>
#include <deque>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include <pybind11/pybind11.h>

namespace py = pybind11;

namespace sw {

// ---------------------------------------------------------------------------
// Wire protocol (little-endian, x86-64 only)
// ---------------------------------------------------------------------------

constexpr uint32_t kMagic = 0x53574159;  // "SWAY"
constexpr uint32_t kProtoVersion = 2;

enum FrameType : uint16_t {
  FT_HELLO = 1,      // payload: PeerInfo blob (both directions on connect)
  FT_EAGER = 2,      // payload: message bytes; hdr.tag/size = message tag/len
  FT_RTS = 3,        // payload: RtsDesc — GPU rendezvous "ready to send"
  FT_RECV_DONE = 4,  // hdr.op_id = sender-side op id whose data was pulled
  FT_RECV_FAIL = 5,  // hdr.op_id + payload = utf-8 error string
  FT_FLUSH_REQ = 6,  // hdr.op_id = flush id; receiver echoes FLUSH_ACK
  FT_FLUSH_ACK = 7,  // hdr.op_id echo
  FT_BYE = 8,        // graceful close notification
  FT_SHM_OFFER = 10,   // payload: u64 ring cap | shm name — same-host only
  FT_SHM_ACK = 11,     // flags: 0 accept / 1 decline; sender's LAST tcp frame
  FT_SHM_SWITCH = 12,  // server's LAST tcp frame after ACK
  FT_RTS_CPU = 13,     // payload: CmaDesc — large same-host CPU rendezvous
  // Small-message inbox plane (smallmsg.hip): sender pushes payloads into
  // the peer's device ring over xGMI; these frames carry only control.
  FT_INBOX_OFFER = 14,   // payload: gpu::InboxInfo of MY ring (peer pushes)
  FT_INBOX_CREDIT = 15,  // hdr.aux = consumed seq (slots <= aux are free)
  FT_SMSG = 16,          // hdr.tag/aux/op_id = tag / size / ring seq
};

#pragma pack(push, 1)
struct FrameHeader {
  uint32_t magic;
  uint16_t type;
  uint16_t flags;
  uint64_t tag;
  uint64_t size;   // payload bytes that follow this header
  uint64_t op_id;  // sender-side op id (RTS/RECV_DONE/FLUSH correlation)
  uint64_t aux;    // type-specific (EAGER: full message length)
};
#pragma pack(pop)
static_assert(sizeof(FrameHeader) == 40);

constexpr int kIpcHandleBytes = 64;  // HIP_IPC_HANDLE_SIZE

namespace gpu {
// Small-message inbox descriptor (smallmsg.hip): a device ring the OWNER
// allocates and exports; the peer pushes payloads into it over xGMI. Sent
// raw as the FT_INBOX_OFFER payload (same-host peers share the ABI).
constexpr uint32_t kInboxHdrBytes = 64;  // [seq][tag][size][pad] per slot

struct InboxInfo {
  uint64_t base = 0;   // device pointer in the OWNER's address space
  int32_t device = -1; // owner's device ordinal
  uint32_t slots = 0;
  uint32_t slot_bytes = 0;  // header + padded payload capacity
  uint8_t handle[kIpcHandleBytes] = {0};
};
}  // namespace gpu

// GPU rendezvous descriptor — carried as RTS payload. The receiver pulls the
// data over xGMI (peer kernel copy) or locally, then acks with RECV_DONE.
#pragma pack(push, 1)
struct RtsDesc {
  uint8_t src_uuid[16];  // sender process uuid — same-process => use raw ptr
  int32_t device;        // sender device ordinal
  uint8_t use_ipc;       // 1 => ipc_handle+offset valid, else raw ptr
  uint8_t pad[3];
  uint8_t ipc_handle[kIpcHandleBytes];
  uint64_t offset;   // byte offset of message start within the ipc allocation
  uint64_t raw_ptr;  // device pointer (same-process fast path)
  // Source layout (0 rows => contiguous): pack happens inside the pull
  // kernel, so non-contiguous tensors move without a .contiguous() pass.
  uint64_t src_rows;
  uint64_t src_row_bytes;
  uint64_t src_stride;
};
#pragma pack(pop)

// Same-host CPU rendezvous descriptor (FT_RTS_CPU payload): the receiver
// pulls straight out of the sender's address space with process_vm_readv
// (the UCX "cma" transport analog) — one copy, out of band of the byte
// stream. The sender keeps the buffer pinned until RECV_DONE.
#pragma pack(push, 1)
struct CmaDesc {
  uint64_t pid;       // sender pid for process_vm_readv
  uint64_t addr;      // source buffer address in the sender
  uint8_t src_uuid[16];
};
#pragma pack(pop)

// PeerInfo blob (HELLO payload / worker-address blob body):
//   u32 version | u64 pid | 16B uuid | u8 has_gpu | i32 gpu_count |
//   u16 name_len | name bytes
struct PeerInfo {
  uint64_t pid = 0;
  uint8_t uuid[16] = {0};
  uint8_t host_id[16] = {0};  // stable per host (boot id + hostname hash)
  bool has_gpu = false;
  int32_t gpu_count = 0;
  std::string name;
};

// Env-tunable knob (config system: STARWAY_* environment variables,
// the analog of the reference's env-driven configuration surface).
inline uint64_t env_u64(const char* name, uint64_t dflt) {
  const char* v = getenv(name);
  if (!v || !*v) return dflt;
  char* end = nullptr;
  unsigned long long x = strtoull(v, &end, 10);
  if (end && (*end == 'k' || *end == 'K')) x <<= 10;
  else if (end && (*end == 'm' || *end == 'M')) x <<= 20;
  else if (end && (*end == 'g' || *end == 'G')) x <<= 30;
  return (uint64_t)x;
}

std::vector<uint8_t> encode_peer_info(const PeerInfo& pi);
bool decode_peer_info(const uint8_t* data, size_t len, PeerInfo* out);
const uint8_t* process_uuid();  // 16 bytes, stable for this process

// ---------------------------------------------------------------------------
// Buffers
// ---------------------------------------------------------------------------

// A caller-provided message buffer. device < 0 => host memory.
// rows > 0 => 2D-strided device buffer (non-contiguous tensor): the
// message is rows x row_bytes dense bytes, row r starting at
// ptr + r*stride; size == rows*row_bytes.
struct BufferRef {
  uint8_t* ptr = nullptr;
  size_t size = 0;
  int device = -1;
  uint64_t rows = 0;
  uint64_t row_bytes = 0;
  uint64_t stride = 0;
};

// Uninitialized heap buffer (std::vector zero-fills on resize, which stalls
// the progress thread for multi-GiB unexpected-message staging).
struct RawBuf {
  std::unique_ptr<uint8_t[]> p;
  size_t n = 0;
  // Returns false on allocation failure (a peer announcing a huge message
  // must not std::terminate the engine thread).
  bool alloc(size_t s) {
    p.reset(new (std::nothrow) uint8_t[s]);
    if (!p && s > 0) {
      n = 0;
      return false;
    }
    n = s;
    return true;
  }
  uint8_t* data() { return p.get(); }
  size_t size() const { return n; }
  bool empty() const { return n == 0; }
  void clear() {
    p.reset();
    n = 0;
  }
};

// ---------------------------------------------------------------------------
// Ops
// ---------------------------------------------------------------------------

enum class OpType : uint8_t { Send, Recv, Flush, FlushEp, Connect };

struct Connection;    // fwd
struct EndpointInfo;  // fwd

struct Op {
  uint64_t id = 0;
  OpType type = OpType::Send;
  BufferRef buf;
  uint64_t tag = 0;
  uint64_t tag_mask = 0;
  Connection* conn = nullptr;  // send target / recv source (once matched)

  // Python state — touched only with the GIL held.
  py::object done_cb;   // Send/Flush: (); Recv: (sender_tag, length)
  py::object fail_cb;   // (reason: str)
  py::object keepalive; // the user's numpy array / torch tensor

  // Keeps the ServerEndpoint alive across the command-queue hop; the engine
  // re-resolves conn from it at processing time.
  std::shared_ptr<EndpointInfo> ep_ref;

  // Progress state (engine thread only).
  uint64_t recv_len = 0;          // actual message length for recv completion
  uint64_t recv_sender_tag = 0;
  bool gpu_send_awaiting_ack = false;
  // Sender-side payload snapshot. Send completion means "buffer reusable":
  // for CMA rendezvous (and eager sends whose bytes could not be written
  // inline) the engine captures the payload here before completing, so a
  // caller that overwrites its buffer right after `await asend` can never
  // corrupt the in-flight message (UCX buffer-reuse contract the reference
  // inherited from ucp_tag_send_nbx).
  RawBuf capture;
  // Op lifetime is owned by the d2h staging list (progress_d2h reaps it);
  // connection-death cleanup must not delete it a second time.
  bool owned_by_d2h = false;
  // Flush bookkeeping (UCX worker/ep-flush semantics: complete when all
  // bytes queued before the flush are written to the wire AND all GPU
  // rendezvous sends posted before it are delivered):
  std::map<Connection*, uint64_t> flush_write_targets;
  std::set<uint64_t> flush_ops_pending;  // GPU send op ids in flight
};

// ---------------------------------------------------------------------------
// Endpoint info (exposed to Python as ServerEndpoint)
// ---------------------------------------------------------------------------

struct EndpointInfo {
  std::string name;
  std::string local_addr;
  int local_port = 0;
  std::string remote_addr;
  int remote_port = 0;
  std::vector<std::pair<std::string, std::string>> transports;
  Connection* conn = nullptr;  // nulled when the connection dies
  class Engine* owner = nullptr;
  // Peer traits snapshotted at accept time (safe to read from Python
  // threads; Connection fields are engine-thread-only).
  bool peer_has_gpu = false;
  bool peer_same_proc = false;
};

// ---------------------------------------------------------------------------
// Connection — one TCP stream + its rx/tx state (engine thread only)
// ---------------------------------------------------------------------------

struct TxItem {
  std::vector<uint8_t> head;      // header (+ inline payload for small frames)
  const uint8_t* ext = nullptr;   // zero-copy user payload (may be null)
  size_t ext_len = 0;
  RawBuf ext_own;                 // owned payload (captured / GPU bounce)
  bool has_keepalive = false;
  bool is_data = false;           // EAGER/RTS frames: droppable on close
  bool via_ring = false;          // carried on the shm ring, not the socket
  bool priority = false;          // control frame inserted ahead of data
  py::object keepalive;           // dropped (under GIL) once fully written
  // Deferred-completion owner: an eager send whose payload could not be
  // captured (allocation failure on a huge message) completes only when its
  // frame is fully written; dropped frames fail the op.
  Op* owner = nullptr;
};

struct UnexpectedMsg {
  uint64_t tag = 0;
  uint64_t size = 0;
  Connection* conn = nullptr;
  bool is_rts = false;
  bool is_cma = false;
  bool is_smsg = false;     // payload parked in our inbox slot
  uint64_t smsg_seq = 0;
  uint64_t sender_op_id = 0;
  RtsDesc rts{};
  CmaDesc cma{};
  RawBuf data;  // eager staging (uninitialized alloc)
  uint64_t staged = 0;  // bytes counted against the conn's unexpected cap
  size_t got = 0;
  bool complete = false;
  Op* bound_recv = nullptr;   // recv matched while message still streaming in
  // When a host recv binds mid-stream, the staged prefix is copied into its
  // buffer and the remaining bytes stream straight there (no double copy).
  uint8_t* redirect_dst = nullptr;
};

class ShmChannel;  // shm.hpp (engine.cpp only)

struct Connection {
  int fd = -1;
  uint64_t conn_id = 0;
  std::string local_addr, remote_addr;
  int local_port = 0, remote_port = 0;
  bool hello_sent = false;
  bool hello_received = false;
  PeerInfo peer;
  bool dead = false;
  std::shared_ptr<EndpointInfo> ep;  // server side

  // --- rx parser ---
  enum class RxState { Header, Payload } rx_state = RxState::Header;
  FrameHeader rx_hdr{};
  size_t rx_got = 0;                 // bytes of current header/payload read
  std::vector<uint8_t> rx_small;     // staging for small control payloads
  // Eager delivery target (exactly one of these while streaming a message):
  Op* rx_recv_op = nullptr;          // matched posted recv (zero-copy fill)
  UnexpectedMsg* rx_unexp = nullptr; // unmatched (staged)
  std::vector<uint8_t> rx_discard;   // truncation sink
  uint64_t rx_msg_remaining = 0;
  bool rx_truncated = false;
  bool rx_discarding = false;  // consume current message into the sink
  // GPU eager-recv host bounce (posted recv buffer is on device):
  RawBuf rx_gpu_bounce;

  // --- shm channel (same-host fast path) ---
  std::unique_ptr<ShmChannel> shm;
  bool shm_tx_enq = false;  // new TxItems go to the ring
  bool shm_rx = false;      // frames are parsed from the ring
  bool sock_eof = false;    // socket closed; conn dies once the ring drains
  bool cma_denied = false;  // peer reported 'cma unavailable': send eager

  // --- flow control ---
  // Receiver side: bytes staged for unmatched (unexpected) eager messages
  // on this connection. Above STARWAY_UNEXP_CAP the engine stops reading
  // the connection, so TCP/ring backpressure reaches the sender instead of
  // this process growing without bound (the UCX receive-window analog).
  uint64_t unexp_staged_bytes = 0;
  bool rx_paused = false;
  // Sender side: rendezvous bytes awaiting RECV_DONE (GPU RTS / CMA / d2h)
  // on this connection; above STARWAY_SEND_WINDOW new sends queue here and
  // start as acks arrive (in-flight window sized for 288 GB HBM3E).
  uint64_t inflight_rndv_bytes = 0;
  std::deque<Op*> deferred_sends;

  // --- small-message inbox (smallmsg.hip) ---
  // Local ring (peer pushes into it):
  gpu::InboxInfo inbox_l;
  bool inbox_l_active = false;
  uint64_t inbox_seen_seq = 0;      // last SMSG sequence processed
  uint64_t inbox_consumed = 0;      // contiguous consumed prefix
  uint64_t inbox_credited = 0;      // last consumed value sent as CREDIT
  std::set<uint64_t> inbox_released;  // out-of-order consumed seqs
  // Remote ring (we push into it):
  gpu::InboxInfo inbox_r;
  bool inbox_r_active = false;
  uint64_t inbox_next_seq = 1;      // next sequence we will assign
  uint64_t inbox_credit_base = 0;   // peer-consumed seq (slot availability)

  // --- tx ---
  std::deque<TxItem> txq;
  size_t tx_front_written = 0;
  uint64_t tx_enqueued_bytes = 0;  // lifetime totals (flush targets)
  uint64_t tx_written_bytes = 0;

  bool want_write() const { return !txq.empty(); }
};

// ---------------------------------------------------------------------------
// Completion record — Python callback to run under the GIL
// ---------------------------------------------------------------------------

struct Completion {
  enum class Kind : uint8_t { SendDone, RecvDone, FlushDone, Fail, Connect,
                              Accept, Close } kind;
  Op* op = nullptr;                  // owned; deleted after firing
  std::string error;                 // Fail / Connect
  uint64_t a = 0, b = 0;             // RecvDone: (sender_tag, length)
  py::object cb0;                    // Connect/Accept/Close callback
  std::shared_ptr<EndpointInfo> ep;  // Accept
};

// ---------------------------------------------------------------------------
// Engine — shared core of Client and Server
// ---------------------------------------------------------------------------

class Context {
 public:
  Context() = default;
  Context(const Context&) = delete;
  Context& operator=(const Context&) = delete;
};

// status: 0 void / 1 init / 2 running / 3 closing / 4 closed
// (state machine contract of reference main.hpp:173-174)
class Engine {
 public:
  enum Mode { ClientMode, ServerMode };

  explicit Engine(Mode mode);
  ~Engine();

  // ---- Python-thread API (GIL held on entry) ----
  void listen(const std::string& addr, int port);     // server, raises
  std::vector<uint8_t> listen_address();              // server worker mode
  void connect(const std::string& addr, int port, py::object cb);
  void connect_address(const std::vector<uint8_t>& blob, py::object cb);
  std::vector<uint8_t> get_worker_address();
  void set_accept_callback(py::object cb);
  void close(py::object cb);

  void send(std::shared_ptr<EndpointInfo> ep, BufferRef buf, uint64_t tag,
            py::object done, py::object fail, py::object keepalive);
  void recv(BufferRef buf, uint64_t tag, uint64_t mask, py::object done,
            py::object fail, py::object keepalive);
  void flush(py::object done, py::object fail);
  void flush_ep(std::shared_ptr<EndpointInfo> ep, py::object done,
                py::object fail);
  std::vector<std::shared_ptr<EndpointInfo>> list_clients();
  double evaluate_perf(std::shared_ptr<EndpointInfo> ep, uint64_t msg_size);

  int status() const { return status_.load(std::memory_order_acquire); }

  // Observability counters (engine thread writes; reads are racy-but-
  // monotonic snapshots, fine for stats).
  struct Stats {
    std::atomic<uint64_t> msgs_sent{0}, msgs_received{0};
    std::atomic<uint64_t> bytes_sent{0}, bytes_received{0};
    std::atomic<uint64_t> eager_rx{0}, gpu_rx{0}, cma_rx{0};
    std::atomic<uint64_t> inbox_rx{0}, inbox_tx{0};  // small-message plane
    std::atomic<uint64_t> doorbell_rx{0};            // of inbox_rx, pre-armed
    std::atomic<uint64_t> unexpected_rx{0};
    std::atomic<uint64_t> unexp_staged_bytes{0};  // flow-control watermark
    std::atomic<uint64_t> deferred_sends{0};      // window-queued sends
  };
  Stats stats_;

 private:
  // ---- engine thread ----
  void thread_main();
  void loop_iteration(bool& did_work);
  void drain_commands(std::vector<Op*>& cmds);
  void process_command(Op* op);
  void start_send(Op* op, Connection* c);   // window-admitted send
  void drain_deferred_sends(Connection* c); // window freed: start queued sends
  void on_send_wire_handoff(Op* op, Connection* c);  // flush retarget helper
  bool cma_eligible(const Op* op, const Connection* c) const;
  void release_window(Op* op);              // rndv send left gpu_sends_
  void unstage_unexp(UnexpectedMsg* um);    // unexpected staging accounting
  void do_connect_start();
  void poll_sockets(int timeout_ms, bool& did_work);
  void handle_readable(Connection* c, bool& did_work);
  void handle_stream(Connection* c, bool from_ring, bool& did_work);
  void handle_writable(Connection* c, bool& did_work);
  void accept_new(bool& did_work);
  void on_frame(Connection* c);           // full header parsed, size==0 or
  void on_frame_payload(Connection* c);   // payload fully staged in rx_small
  void begin_eager(Connection* c);
  void finish_eager_into_recv(Connection* c);
  void handle_rts(Connection* c, const RtsDesc& rts, uint64_t tag,
                  uint64_t size, uint64_t sender_op);
  void start_gpu_pull(Op* recv_op, const RtsDesc& rts, uint64_t tag,
                      uint64_t size, uint64_t sender_op, Connection* c);
  void start_cma_pull(Op* recv_op, const CmaDesc& cma, uint64_t tag,
                      uint64_t size, uint64_t sender_op, Connection* c);
  void progress_cma(bool& did_work);
  void match_or_stash_recv(Op* op);
  bool try_match_unexpected(Op* op);
  void complete_recv_from_unexpected(Op* op, UnexpectedMsg* um);
  void poll_gpu(bool& did_work);
  void enqueue_frame(Connection* c, FrameType t, uint64_t tag, uint64_t op_id,
                     uint64_t aux, const void* payload, size_t payload_len,
                     bool priority, uint16_t flags = 0);
  // Returns true when the payload is captured (written out or copied into
  // engine-owned memory) => the op may complete immediately. False => the
  // TxItem owns the op (deferred completion at write-out).
  bool enqueue_eager(Connection* c, Op* op);
  void send_hello(Connection* c);
  void on_hello(Connection* c);
  void on_conn_dead(Connection* c);
  void check_flush_progress();
  void on_gpu_send_acked(uint64_t op_id, bool failed, const std::string& err);
  void teardown();
  void fire_completions();  // acquires GIL, drains completions_
  void complete(Completion&& comp);
  void fail_op(Op* op, const std::string& reason);
  void wake();
  Connection* make_listener(const std::string& addr, int port);

  double perf_model(bool peer_gpu, bool same_proc, uint64_t msg_size) const;
  // Client-side peer traits (bit0 has_gpu, bit1 same_proc; -1 unknown),
  // published at HELLO so evaluate_perf never touches engine-owned state.
  std::atomic<int> client_peer_traits_{-1};

 public:
  Mode mode_;
  std::atomic<int> status_{0};
  // HIP device the inbox ring (and other engine-owned device state) lands
  // on: captured from the CONSTRUCTING thread's current device (the
  // engine's own thread never calls hipSetDevice). -1 = no GPU.
  int preferred_device_ = -1;
  // Per-engine stream lane for the small-message kernels: engines in one
  // process must not share a doorbell (Wait) stream.
  int engine_lane_ = 0;

 private:
  // Command queue: Python threads -> engine thread.
  std::mutex cmd_mu_;
  std::vector<Op*> cmd_queue_;
  std::atomic<bool> cmd_pending_{false};
  // True while the engine loop is in hot-spin mode (poll timeout 0):
  // submitters skip the wake-pipe write, saving a syscall per op.
  std::atomic<bool> engine_hot_{true};
  int wake_fds_[2] = {-1, -1};  // self-pipe to interrupt poll()

  std::thread thread_;
  std::atomic<uint64_t> next_op_id_{1};

  // Engine-thread state.
  int listen_fd_ = -1;
  std::string listen_host_;
  int listen_port_ = 0;
  bool worker_mode_ = false;
  std::vector<std::unique_ptr<Connection>> conns_;
  uint64_t next_conn_id_ = 1;
  std::deque<Op*> posted_recvs_;
  std::deque<std::unique_ptr<UnexpectedMsg>> unexpected_;
  std::vector<Op*> pending_flushes_;
  std::unordered_map<uint64_t, Op*> gpu_sends_;  // op id -> awaiting ack
  struct GpuPull;  // defined in engine.cpp (holds hipEvent)
  std::vector<std::unique_ptr<GpuPull>> gpu_pulls_;
  // Small device pulls collected during a drain, launched batched at the
  // end of the loop iteration (one kernel + one event for up to 8).
  struct SmallPull {
    RtsDesc rts;
    Op* recv_op;
    Connection* conn;
    uint64_t sender_op_id;
    uint64_t tag;
    uint64_t size;
  };
  std::vector<SmallPull> pending_small_pulls_;
  void flush_small_pulls();
  // Cross-host GPU send: device->host bounce in flight; when the copy
  // completes the payload goes out as a plain eager frame.
  struct D2hSend {
    void* ticket = nullptr;
    Op* op = nullptr;
    RawBuf buf;
  };
  std::vector<std::unique_ptr<D2hSend>> d2h_sends_;
  void progress_d2h(bool& did_work);
  struct CmaPull {
    Op* recv_op = nullptr;
    Connection* conn = nullptr;
    uint64_t sender_op_id = 0;
    uint64_t tag = 0;
    uint64_t size = 0;
    uint64_t done = 0;
    CmaDesc desc{};
    RawBuf bounce;  // host staging when the posted recv buffer is on device
  };
  std::vector<std::unique_ptr<CmaPull>> cma_pulls_;

  // --- small-message inbox machinery (engine thread only) ---
  struct PendingPush {
    Op* op;
    uint64_t seq;
  };
  std::map<Connection*, std::vector<PendingPush>> pending_pushes_;
  struct PushBatch {
    void* ticket = nullptr;
    Connection* conn = nullptr;
    std::vector<Op*> ops;
  };
  std::vector<std::unique_ptr<PushBatch>> push_batches_;
  struct PendingUnpack {
    Connection* conn;
    uint64_t seq;
    uint64_t size;
    uint64_t tag;
    Op* recv_op;
    uint8_t* dst;  // device dst, or null => pinned bounce
    int retries = 0;
  };
  std::vector<PendingUnpack> pending_unpacks_;
  struct UnpackBatch {
    void* ticket = nullptr;
    Connection* conn = nullptr;
    std::vector<PendingUnpack> msgs;
    std::vector<bool> done;
    size_t remaining = 0;
    bool probe = false;  // inbox bring-up probe: no recv op attached
  };
  std::vector<std::unique_ptr<UnpackBatch>> unpack_batches_;
  // Doorbell: at most one pre-armed wait kernel per engine.
  struct Armed {
    void* ticket = nullptr;
    Op* recv_op = nullptr;
    Connection* conn = nullptr;
    uint64_t seq = 0;
  } armed_;
  // Doorbell re-arm backoff: after a few consecutive unused expiries the
  // engine stops relaunching until inbox traffic resumes or a recv is
  // posted — back-to-back spinning doorbells monopolize their hardware
  // queue and starve co-mapped streams under queue oversubscription.
  int arm_streak_ = 0;
  bool arm_backoff_ = false;
  // Canceled doorbells whose kernel had not reported by the cancel wait:
  // polled until they resolve so their pinned cell is never reused while
  // a late-starting kernel could still write it.
  std::vector<void*> zombie_arms_;
  void retire_armed_ticket(void* t);  // free now or park as zombie
  void poll_zombie_arms();
  // A doorbell copy whose SMSG control frame has not arrived yet (the
  // kernel consumed the message during a disarm race).
  struct ArmedDone {
    bool active = false;
    Connection* conn = nullptr;
    uint64_t seq = 0;
    uint64_t size = 0;
    Op* recv_op = nullptr;
  } armed_done_;
  void handle_smsg(Connection* c, uint64_t tag, uint64_t size, uint64_t seq);
  void dispatch_smsg_to_recv(Connection* c, Op* r, uint64_t tag,
                             uint64_t size, uint64_t seq);
  void flush_pending_pushes();
  void progress_pushes(bool& did_work);
  void flush_pending_unpacks();
  void progress_unpacks(bool& did_work);
  void progress_armed(bool& did_work);
  void inbox_release_seq(Connection* c, uint64_t seq);
  void try_arm();
  bool steal_armed(Op* r);  // false => the doorbell consumed r
  Op* take_matching_recv(uint64_t tag);
  void repost_recv_front(Op* r);  // disarms, then push_front

  // Connect state (client).
  std::string connect_host_;
  int connect_port_ = 0;
  // Worker-address mode: fallback routes tried in order.
  std::vector<std::pair<std::string, int>> connect_candidates_;
  py::object connect_cb_;
  bool connect_requested_ = false;

  py::object accept_cb_;
  std::mutex close_mu_;  // orders close_cb_ assignment vs teardown's read
  py::object close_cb_;
  bool have_accept_cb_ = false;

  std::vector<Completion> completions_;
  // py::objects whose final decref must wait for a GIL hold (moved here from
  // engine-thread contexts; cleared inside fire_completions).
  std::vector<py::object> dead_objs_;

  // Endpoint registry (shared with Python thread under ep_mu_).
  std::mutex ep_mu_;
  std::vector<std::shared_ptr<EndpointInfo>> eps_;

  uint64_t idle_iters_ = 0;
};

// ---------------------------------------------------------------------------
// GPU layer (gpu.cpp) — all HIP calls live behind this interface so the
// engine compiles and runs on GPU-less hosts.
// ---------------------------------------------------------------------------

namespace gpu {
bool available();
int device_count();
int current_device();  // calling thread's device, -1 without GPU
// Fill an RtsDesc for a device buffer (ipc handle or raw ptr).
bool make_rts(const BufferRef& buf, RtsDesc* out, std::string* err);
// Begin an async pull of `size` bytes described by `rts` into dst (host or
// device). Returns an opaque ticket (hipEvent) or null on error.
void* begin_pull(const RtsDesc& rts, const BufferRef& dst, uint64_t size,
                 std::string* err);
// Batched small-message pull: one launch + one event for up to 8 messages
// targeting the same destination device. Each entry must be a contiguous
// device->device transfer. Returns one shared ticket.
struct PullReq {
  RtsDesc rts;
  uint8_t* dst_ptr;
  int dst_device;
  uint64_t size;
};
void* begin_pull_multi(const PullReq* reqs, int n, std::string* err);
// Begin an async host->device upload into a device buffer. The host memory
// must stay valid until the ticket completes; hand its ownership to the
// ticket with attach_bounce.
void* begin_h2d(const BufferRef& dst, const void* src, uint64_t size,
                std::string* err);
// Device -> host staging download (cross-host GPU sends).
void* begin_d2h(void* host_dst, const BufferRef& src, std::string* err);
void attach_bounce(void* ticket, RawBuf&& bounce);
// IPC hygiene: close every imported hipIpc mapping and drop both handle
// caches (process-exit hook + explicit invalidation for allocators that
// return memory to the driver, e.g. torch empty_cache).
void ipc_close_all();

// ---------------------------------------------------------------------------
// Small-message inbox data plane (smallmsg.hip): sender-push ring in the
// receiver's HBM, written over xGMI; local unpack or pre-armed doorbell
// kernel on the receiver. See smallmsg.hip header comment for the design;
// InboxInfo / kInboxHdrBytes are declared above (wire ABI).
// ---------------------------------------------------------------------------
bool inbox_create(InboxInfo* out, int device, std::string* err);
void inbox_destroy(const InboxInfo& ib);

struct PushMsg {
  const uint8_t* src;
  uint32_t size;
  uint64_t seq;
  uint64_t tag;
};
// Batched push into the PEER's inbox; returns an event ticket polled with
// push_poll / freed with push_free.
void* inbox_push(const InboxInfo& peer, bool same_proc, int run_device,
                 const PushMsg* msgs, int n, int lane, std::string* err);
int push_poll(void* ticket, std::string* err);
void push_free(void* ticket);

struct UnpackMsg {
  uint64_t seq;
  uint32_t size;
  uint8_t* dst;  // device pointer, or null => pinned host bounce
};
void* inbox_unpack(const InboxInfo& mine, const UnpackMsg* msgs, int n,
                   int lane, std::string* err);
int unpack_poll(void* ticket, int idx, std::string* err);  // 0/1/-1
const uint8_t* unpack_bounce(void* ticket, int idx);
void unpack_free(void* ticket);

// Doorbell: pre-armed bounded wait for the next sequence number, copying
// into dst on a tag match. arm_poll: 0 running / 1 copied / 2 nomatch /
// 3 canceled / 4 expired.
void* arm_recv(const InboxInfo& mine, uint64_t expect_seq, uint64_t tag,
               uint64_t mask, uint8_t* dst, uint64_t max_size, int lane,
               std::string* err);
int arm_poll(void* ticket, uint64_t* size_out);
void arm_cancel(void* ticket);
void arm_free(void* ticket);
void arm_leak(void* ticket);  // kernel may still write: do not reuse cell
// Poll a ticket: 1 done, 0 pending, -1 error.
int poll_ticket(void* ticket, std::string* err);
void free_ticket(void* ticket);
void synchronize_all();
bool copy_device_sync(void* dst, const void* src, size_t n, int device,
                      std::string* err);
// Stats for evaluate_perf / transports introspection. Values come from a
// cached on-box copy microbenchmark (~/.cache/starway/perf.cal) with
// round-1 measurements as analytic fallback.
double same_gpu_copy_gbps();
double xgmi_link_gbps();
bool calibrate(bool force, std::string* err);  // run + persist microbench
}  // namespace gpu

}  // namespace sw
