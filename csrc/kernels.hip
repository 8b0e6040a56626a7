// gfx950 (CDNA4 / MI355X) data-movement kernels for starway_amd.
//
// These are the hand-written copy kernels that replace what UCX's internal
// copy paths did for the reference (ucp_tag_send/recv delivery). A tagged
// message delivery is one device-side copy: local HBM->HBM on one GPU, or a
// peer read over xGMI when src is another GPU's memory (hipIpc-mapped or
// same-process peer pointer).
//
// Design (per MI355X microarch):
//  * 64-wide wavefronts; 256-thread workgroups (4 waves)
//  * 16 B/lane vector moves (global_load_dwordx4 / global_store_dwordx4):
//    1 KiB per wave-instruction — the measured-fastest HBM streaming shape
//    (float4 copy reaches 6.29 TB/s read on this chip)
//  * grid-stride loop sized >> 256 workgroups so all 8 XCDs fill
//  * unroll by 4 so each thread has 4 independent loads in flight (latency
//    hiding without LDS staging — a pure stream copy has no reuse, so LDS
//    would only add a round trip)
//  * non-temporal variants for large messages: a message buffer is read and
//    written exactly once, so polluting L2/LLC with it costs other traffic.
#include <hip/hip_runtime.h>

#include <cstdint>

namespace sw {

// ---------------------------------------------------------------------------
// 16B-vector bulk copy (both pointers 16B-aligned), grid-stride, x4 unroll.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_copy_b128(
    const uint4* __restrict__ src, uint4* __restrict__ dst, size_t n16) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  // 4 independent elements in flight per thread per iteration.
  while (i + 3 * stride < n16) {
    uint4 a = src[i];
    uint4 b = src[i + stride];
    uint4 c = src[i + 2 * stride];
    uint4 d = src[i + 3 * stride];
    dst[i] = a;
    dst[i + stride] = b;
    dst[i + 2 * stride] = c;
    dst[i + 3 * stride] = d;
    i += 4 * stride;
  }
  for (; i < n16; i += stride) dst[i] = src[i];
}

// Non-temporal variant: bypass-cache hints on both sides. Used for messages
// past the LLC-thrash threshold (they are touched exactly once). The builtin
// wants a native vector type, not HIP_vector_type.
using u32x4 = __attribute__((ext_vector_type(4))) unsigned int;

__global__ __launch_bounds__(256) void k_copy_b128_nt(
    const u32x4* __restrict__ src, u32x4* __restrict__ dst, size_t n16) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  while (i + 3 * stride < n16) {
    u32x4 a = __builtin_nontemporal_load(&src[i]);
    u32x4 b = __builtin_nontemporal_load(&src[i + stride]);
    u32x4 c = __builtin_nontemporal_load(&src[i + 2 * stride]);
    u32x4 d = __builtin_nontemporal_load(&src[i + 3 * stride]);
    __builtin_nontemporal_store(a, &dst[i]);
    __builtin_nontemporal_store(b, &dst[i + stride]);
    __builtin_nontemporal_store(c, &dst[i + 2 * stride]);
    __builtin_nontemporal_store(d, &dst[i + 3 * stride]);
    i += 4 * stride;
  }
  for (; i < n16; i += stride) {
    u32x4 a = __builtin_nontemporal_load(&src[i]);
    __builtin_nontemporal_store(a, &dst[i]);
  }
}

// Deeper-unroll NT variant (8 independent 16B elements in flight/thread).
__global__ __launch_bounds__(256) void k_copy_b128_nt_u8(
    const u32x4* __restrict__ src, u32x4* __restrict__ dst, size_t n16) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  while (i + 7 * stride < n16) {
    u32x4 v0 = __builtin_nontemporal_load(&src[i]);
    u32x4 v1 = __builtin_nontemporal_load(&src[i + stride]);
    u32x4 v2 = __builtin_nontemporal_load(&src[i + 2 * stride]);
    u32x4 v3 = __builtin_nontemporal_load(&src[i + 3 * stride]);
    u32x4 v4 = __builtin_nontemporal_load(&src[i + 4 * stride]);
    u32x4 v5 = __builtin_nontemporal_load(&src[i + 5 * stride]);
    u32x4 v6 = __builtin_nontemporal_load(&src[i + 6 * stride]);
    u32x4 v7 = __builtin_nontemporal_load(&src[i + 7 * stride]);
    __builtin_nontemporal_store(v0, &dst[i]);
    __builtin_nontemporal_store(v1, &dst[i + stride]);
    __builtin_nontemporal_store(v2, &dst[i + 2 * stride]);
    __builtin_nontemporal_store(v3, &dst[i + 3 * stride]);
    __builtin_nontemporal_store(v4, &dst[i + 4 * stride]);
    __builtin_nontemporal_store(v5, &dst[i + 5 * stride]);
    __builtin_nontemporal_store(v6, &dst[i + 6 * stride]);
    __builtin_nontemporal_store(v7, &dst[i + 7 * stride]);
    i += 8 * stride;
  }
  for (; i < n16; i += stride) {
    u32x4 a = __builtin_nontemporal_load(&src[i]);
    __builtin_nontemporal_store(a, &dst[i]);
  }
}

// Byte-granularity fallback for arbitrary (mis)alignment.
__global__ __launch_bounds__(256) void k_copy_b8(
    const uint8_t* __restrict__ src, uint8_t* __restrict__ dst, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = src[i];
}

// dword fallback when both sides are 4B-co-aligned but not 16B.
__global__ __launch_bounds__(256) void k_copy_b32(
    const uint32_t* __restrict__ src, uint32_t* __restrict__ dst, size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) dst[i] = src[i];
}

// ---------------------------------------------------------------------------
// Strided (2D) pack/unpack copy: row r of the message lives at
// src + r*src_stride and lands at dst + r*dst_stride; rows are dense
// row_bytes-long runs. Used for non-contiguous device tensors (e.g. torch
// slices) so they move without a .contiguous() staging pass. Lanes walk
// 16B elements in row-major message order, so global accesses stay
// coalesced within each row.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_copy_strided_b128(
    const uint8_t* __restrict__ src, uint64_t src_stride,
    uint8_t* __restrict__ dst, uint64_t dst_stride, uint64_t rows,
    uint64_t row_n16) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t total = (size_t)rows * row_n16;
  for (; i < total; i += stride) {
    size_t r = i / row_n16;
    size_t c = i - r * row_n16;
    *(uint4*)(dst + r * dst_stride + c * 16) =
        *(const uint4*)(src + r * src_stride + c * 16);
  }
}

__global__ __launch_bounds__(256) void k_copy_strided_b8(
    const uint8_t* __restrict__ src, uint64_t src_stride,
    uint8_t* __restrict__ dst, uint64_t dst_stride, uint64_t rows,
    uint64_t row_bytes) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t total = (size_t)rows * row_bytes;
  for (; i < total; i += stride) {
    size_t r = i / row_bytes;
    size_t c = i - r * row_bytes;
    dst[r * dst_stride + c] = src[r * src_stride + c];
  }
}

// ---------------------------------------------------------------------------
// Batched small-message copy: up to kMultiMax messages per launch, one
// block per message (messages <= ~64 KiB). Amortizes the launch + event
// cost that dominates small-message rate (measured: raw engine pipeline
// sustains 600k msgs/s on host buffers but 85k with one launch+event per
// device message).
// ---------------------------------------------------------------------------
struct MultiCopyDesc {
  const uint8_t* src;
  uint8_t* dst;
  uint32_t bytes;
};
constexpr int kMultiMax = 8;
struct MultiCopyArgs {
  MultiCopyDesc d[kMultiMax];
  int n;
};

__global__ __launch_bounds__(256) void k_copy_multi(MultiCopyArgs args) {
  const MultiCopyDesc& m = args.d[blockIdx.x];
  uintptr_t sp = (uintptr_t)m.src, dp = (uintptr_t)m.dst;
  uint32_t bytes = m.bytes;
  if ((sp & 15) == 0 && (dp & 15) == 0 && (bytes & 15) == 0) {
    const uint4* s4 = (const uint4*)sp;
    uint4* d4 = (uint4*)dp;
    for (uint32_t i = threadIdx.x; i < bytes / 16; i += blockDim.x)
      d4[i] = s4[i];
  } else {
    const uint8_t* s1 = (const uint8_t*)sp;
    uint8_t* d1 = (uint8_t*)dp;
    for (uint32_t i = threadIdx.x; i < bytes; i += blockDim.x)
      d1[i] = s1[i];
  }
}

hipError_t launch_copy_multi(const MultiCopyDesc* descs, int n,
                             hipStream_t stream) {
  MultiCopyArgs args;
  for (int i = 0; i < n; i++) args.d[i] = descs[i];
  args.n = n;
  hipLaunchKernelGGL(k_copy_multi, dim3(n), dim3(256), 0, stream, args);
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// Host-side dispatch
// ---------------------------------------------------------------------------

static inline int copy_grid(size_t work_items) {
  // >> 256 WGs to fill 8 XCDs x 32 CUs; cap so tiny copies stay one-wave-ish.
  // Cap tunable via STARWAY_COPY_BLOCKS for on-box sweeps.
  static const size_t cap = [] {
    const char* v = getenv("STARWAY_COPY_BLOCKS");
    return v && *v ? strtoull(v, nullptr, 10) : 8192ull;
  }();
  size_t blocks = (work_items + 255) / 256;
  if (blocks < 1) blocks = 1;
  if (blocks > cap) blocks = cap;
  return (int)blocks;
}

hipError_t launch_copy(void* dst, const void* src, size_t bytes,
                       hipStream_t stream) {
  if (bytes == 0) return hipSuccess;
  uintptr_t s = (uintptr_t)src, d = (uintptr_t)dst;
  if ((s & 15) == (d & 15)) {
    // Co-aligned: byte head to the next 16B boundary, vector bulk, byte tail.
    size_t head = (16 - (s & 15)) & 15;
    if (head > bytes) head = bytes;
    if (head) {
      hipLaunchKernelGGL(k_copy_b8, dim3(1), dim3(64), 0, stream,
                         (const uint8_t*)src, (uint8_t*)dst, head);
      s += head;
      d += head;
      bytes -= head;
    }
    size_t n16 = bytes / 16;
    size_t tail = bytes & 15;
    if (n16) {
      // NT past 64 MiB (STARWAY_NT_THRESHOLD): the copy would otherwise
      // sweep the 256 MiB LLC.
      static const size_t kNtThreshold = [] {
        const char* v = getenv("STARWAY_NT_THRESHOLD");
        return v && *v ? (size_t)strtoull(v, nullptr, 10) : (size_t)64 << 20;
      }();
      static const int nt_unroll = [] {
        const char* v = getenv("STARWAY_COPY_UNROLL");
        return v && *v ? atoi(v) : 4;
      }();
      if (bytes >= kNtThreshold) {
        if (nt_unroll >= 8) {
          hipLaunchKernelGGL(k_copy_b128_nt_u8, dim3(copy_grid(n16 / 8 + 1)),
                             dim3(256), 0, stream, (const u32x4*)s,
                             (u32x4*)d, n16);
        } else {
          hipLaunchKernelGGL(k_copy_b128_nt, dim3(copy_grid(n16 / 4 + 1)),
                             dim3(256), 0, stream, (const u32x4*)s,
                             (u32x4*)d, n16);
        }
      } else {
        hipLaunchKernelGGL(k_copy_b128, dim3(copy_grid(n16 / 4 + 1)),
                           dim3(256), 0, stream, (const uint4*)s, (uint4*)d,
                           n16);
      }
    }
    if (tail) {
      hipLaunchKernelGGL(k_copy_b8, dim3(1), dim3(64), 0, stream,
                         (const uint8_t*)(s + n16 * 16),
                         (uint8_t*)(d + n16 * 16), tail);
    }
  } else if ((s & 3) == (d & 3) && (s & 3) == 0 && (bytes & 3) == 0) {
    size_t n4 = bytes / 4;
    hipLaunchKernelGGL(k_copy_b32, dim3(copy_grid(n4 / 4 + 1)), dim3(256), 0,
                       stream, (const uint32_t*)src, (uint32_t*)dst, n4);
  } else {
    hipLaunchKernelGGL(k_copy_b8, dim3(copy_grid(bytes / 16 + 1)), dim3(256),
                       0, stream, (const uint8_t*)src, (uint8_t*)dst, bytes);
  }
  return hipGetLastError();
}

hipError_t launch_copy_strided(void* dst, uint64_t dst_stride,
                               const void* src, uint64_t src_stride,
                               uint64_t rows, uint64_t row_bytes,
                               hipStream_t stream) {
  if (rows == 0 || row_bytes == 0) return hipSuccess;
  uintptr_t s = (uintptr_t)src, d = (uintptr_t)dst;
  bool vec = (s & 15) == 0 && (d & 15) == 0 && (src_stride & 15) == 0 &&
             (dst_stride & 15) == 0 && (row_bytes & 15) == 0;
  if (vec) {
    uint64_t row_n16 = row_bytes / 16;
    size_t total = (size_t)rows * row_n16;
    hipLaunchKernelGGL(k_copy_strided_b128, dim3(copy_grid(total / 4 + 1)),
                       dim3(256), 0, stream, (const uint8_t*)src, src_stride,
                       (uint8_t*)dst, dst_stride, rows, row_n16);
  } else {
    size_t total = (size_t)rows * row_bytes;
    hipLaunchKernelGGL(k_copy_strided_b8, dim3(copy_grid(total / 16 + 1)),
                       dim3(256), 0, stream, (const uint8_t*)src, src_stride,
                       (uint8_t*)dst, dst_stride, rows, row_bytes);
  }
  return hipGetLastError();
}

}  // namespace sw
