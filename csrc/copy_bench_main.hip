// Standalone copy-kernel microbench (no Python/torch) — used for rocprofv3
// PMC counter collection, where tracing the full Python process has been
// seen to crash the profiler.  Usage: copy_bench [bytes] [iters]
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

namespace sw {
hipError_t launch_copy(void* dst, const void* src, size_t bytes,
                       hipStream_t stream);
}

#define CHECK(x)                                                    \
  do {                                                              \
    hipError_t e_ = (x);                                            \
    if (e_ != hipSuccess) {                                         \
      fprintf(stderr, "%s failed: %s\n", #x, hipGetErrorString(e_)); \
      return 1;                                                     \
    }                                                               \
  } while (0)

int main(int argc, char** argv) {
  size_t bytes = argc > 1 ? strtoull(argv[1], nullptr, 10) : (256ull << 20);
  int iters = argc > 2 ? atoi(argv[2]) : 10;
  void *src, *dst;
  CHECK(hipMalloc(&src, bytes));
  CHECK(hipMalloc(&dst, bytes));
  CHECK(hipMemset(src, 0x5A, bytes));
  hipStream_t stream;
  CHECK(hipStreamCreate(&stream));
  // warmup + correctness spot check
  CHECK(sw::launch_copy(dst, src, bytes, stream));
  CHECK(hipStreamSynchronize(stream));
  unsigned char probe[16];
  CHECK(hipMemcpy(probe, (char*)dst + bytes / 2, 16, hipMemcpyDeviceToHost));
  for (int i = 0; i < 16; i++)
    if (probe[i] != 0x5A) {
      fprintf(stderr, "copy mismatch at probe %d\n", i);
      return 1;
    }
  hipEvent_t t0, t1;
  CHECK(hipEventCreate(&t0));
  CHECK(hipEventCreate(&t1));
  CHECK(hipEventRecord(t0, stream));
  for (int i = 0; i < iters; i++)
    CHECK(sw::launch_copy(dst, src, bytes, stream));
  CHECK(hipEventRecord(t1, stream));
  CHECK(hipStreamSynchronize(stream));
  float ms = 0;
  CHECK(hipEventElapsedTime(&ms, t0, t1));
  double s = ms / 1e3 / iters;
  printf("%zu bytes: %.3f TB/s payload (%.3f TB/s HBM traffic), %.1f us\n",
         bytes, bytes / s / 1e12, 2 * bytes / s / 1e12, s * 1e6);
  return 0;
}
