// GPU-less stub of the gpu:: interface for sanitizer builds (TSAN/ASAN
// stress binary, csrc/tsan_stress_main.cpp). The engine's cross-thread
// contracts are identical with or without a device, so the host transport
// paths (TCP, shm ring, CMA, tag matching, flush, teardown) run fully
// instrumented without linking HIP.
#include "core.hpp"

namespace sw {
namespace gpu {

bool available() { return false; }
int device_count() { return 0; }
int current_device() { return -1; }
int device_of(const void*) { return -1; }

bool make_rts(const BufferRef&, RtsDesc*, std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return false;
}
void* begin_pull(const RtsDesc&, const BufferRef&, uint64_t,
                 std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return nullptr;
}
void* begin_pull_multi(const PullReq*, int, std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return nullptr;
}
void* begin_h2d(const BufferRef&, const void*, uint64_t, std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return nullptr;
}
void* begin_d2h(void*, const BufferRef&, std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return nullptr;
}
void attach_bounce(void*, RawBuf&&) {}
int poll_ticket(void*, std::string*) { return -1; }
void free_ticket(void*) {}
void synchronize_all() {}
bool copy_device_sync(void*, const void*, size_t, int, std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return false;
}
double same_gpu_copy_gbps() { return 3100.0; }
double xgmi_link_gbps() { return 140.0; }
bool calibrate(bool, std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return false;
}
void ipc_close_all() {}

bool inbox_create(InboxInfo*, int, std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return false;
}
void inbox_destroy(const InboxInfo&) {}
void* inbox_push(const InboxInfo&, bool, int, const PushMsg*, int, int,
                 std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return nullptr;
}
int push_poll(void*, std::string*) { return -1; }
void push_free(void*) {}
void* inbox_unpack(const InboxInfo&, const UnpackMsg*, int, int,
                   std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return nullptr;
}
int unpack_poll(void*, int, std::string*) { return -1; }
const uint8_t* unpack_bounce(void*, int) { return nullptr; }
void unpack_free(void*) {}
void* arm_recv(const InboxInfo&, uint64_t, uint64_t, uint64_t, uint8_t*,
               uint64_t, int, std::string* err) {
  *err = "no GPU (sanitizer stub)";
  return nullptr;
}
int arm_poll(void*, uint64_t*) { return 4; }
void arm_cancel(void*) {}
void arm_free(void*) {}

}  // namespace gpu
}  // namespace sw

namespace sw {
namespace gpu {
void arm_leak(void*) {}
}  // namespace gpu
}  // namespace sw
