// grpcoll.hip — native group-collective transport over HIP IPC + xGMI
// (gfx950). MI355X-first redesign of the reference's DeepEP-style grpcoll
// (csrc/comm/grpcoll/kernels/intranode_kernel.cuh:46 group_cast_kernel,
// :573 group_reduce_kernel — NVLink-IPC peer buffers, SM-pair channels,
// head/tail token queues).
//
// Design here is PULL-based, which suits xGMI's point-to-point links and
// needs no channel queues: every rank exposes a persistent window buffer
// via hipIpcGetMemHandle (dmabuf mode); a producer COPIES its payload into
// its own window (SDMA, off the CUs) and a 1-block signal kernel publishes
// a monotonically increasing sequence number with a system-scope release.
// Consumers run ONE bounded-grid pull kernel that (a) spin-waits each
// source peer's flag (system-scope acquire, s_sleep between polls), then
// (b) gathers its planned row ranges straight out of the peers' windows
// over xGMI into the local stage buffer — or sum-reduces them into the
// local accumulator (fp32) for the backward partial-dKV return. All
// ordering is device-side: no host sync between compute and comm.
#include <hip/hip_runtime.h>

#include "../../include/magi_ffa.h"

#define GRPCOLL_MAX_PEERS 8

extern "C" int magi_ipc_get_handle(const void* dev_ptr, void* out_handle) {
  return (int)hipIpcGetMemHandle((hipIpcMemHandle_t*)out_handle,
                                 const_cast<void*>(dev_ptr));
}

extern "C" int magi_ipc_open(const void* handle, void** out_ptr) {
  return (int)hipIpcOpenMemHandle(out_ptr, *(const hipIpcMemHandle_t*)handle,
                                  hipIpcMemLazyEnablePeerAccess);
}

extern "C" int magi_ipc_close(void* ptr) {
  return (int)hipIpcCloseMemHandle(ptr);
}

// ---- signal: publish seq into this rank's flag slot (release) ----
__global__ void grpcoll_signal_kernel(int* flag, int value) {
  if (threadIdx.x == 0)
    __hip_atomic_store(flag, value, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_SYSTEM);
}

extern "C" int magi_grpcoll_signal(void* flag_ptr, int value, void* stream) {
  hipLaunchKernelGGL(grpcoll_signal_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (int*)flag_ptr, value);
  return (int)hipGetLastError();
}

// piece table entry (built at plan time, device int32 tensor [n, 4]):
//   {peer_idx, src_row (in the peer's window), dst_row, n_rows}
struct GrpCollPullArgs {
  const int* pieces;     // [n_pieces * 4]
  int n_pieces;
  int row_elems;         // elements per row (h*d)
  int elem_size;         // bytes per element of the payload
  const void* peer_ptrs[GRPCOLL_MAX_PEERS];  // peers' window base pointers
  const int* peer_flags[GRPCOLL_MAX_PEERS];  // peers' flag slots
  int wait_value;        // spin until every used peer's flag >= this
  int n_peers;
  void* dst;             // stage buffer (cast) / local accumulator (reduce)
  int reduce;            // 0 = copy, 1 = fp32 sum
};

// bounded-grid pull: block b walks pieces b, b+G, ... — 16-byte lanes.
// The grid is capped (<=64 WGs) so spinning never starves the producers'
// copy/signal work (copies ride the SDMA engines anyway).
__global__ __launch_bounds__(256) void grpcoll_pull_kernel(GrpCollPullArgs a) {
  // one thread per block polls each used peer's flag, then the block syncs
  __shared__ int ready;
  if (threadIdx.x == 0) {
    for (int p = 0; p < a.n_peers; ++p) {
      if (!a.peer_flags[p]) continue;
      while (__hip_atomic_load(a.peer_flags[p], __ATOMIC_ACQUIRE,
                               __HIP_MEMORY_SCOPE_SYSTEM) < a.wait_value) {
        __builtin_amdgcn_s_sleep(32);
      }
    }
    ready = 1;
  }
  __syncthreads();
  (void)ready;

  const long long row_bytes = (long long)a.row_elems * a.elem_size;
  for (int pi = blockIdx.x; pi < a.n_pieces; pi += gridDim.x) {
    const int peer = a.pieces[4 * pi];
    const int src_row = a.pieces[4 * pi + 1];
    const int dst_row = a.pieces[4 * pi + 2];
    const int n_rows = a.pieces[4 * pi + 3];
    const char* src = (const char*)a.peer_ptrs[peer] + src_row * row_bytes;
    char* dst = (char*)a.dst + dst_row * row_bytes;
    const long long total16 = n_rows * row_bytes / 16;
    if (!a.reduce) {
      for (long long i = threadIdx.x; i < total16; i += blockDim.x) {
        ((float4*)dst)[i] = ((const float4*)src)[i];
      }
    } else {
      // fp32 sum into the local accumulator (rows are fp32 here)
      for (long long i = threadIdx.x; i < total16; i += blockDim.x) {
        float4 s4 = ((const float4*)src)[i];
        float4* d4 = (float4*)dst + i;
        float4 d = *d4;
        d.x += s4.x; d.y += s4.y; d.z += s4.z; d.w += s4.w;
        *d4 = d;
      }
    }
  }
}

struct magi_grpcoll_pull_args {
  const int* pieces;
  int n_pieces;
  int row_elems;
  int elem_size;
  const void* peer_ptrs[GRPCOLL_MAX_PEERS];
  const int* peer_flags[GRPCOLL_MAX_PEERS];
  int wait_value;
  int n_peers;
  void* dst;
  int reduce;
  void* stream;
};

extern "C" int magi_grpcoll_pull(const magi_grpcoll_pull_args* a) {
  if (!a || a->n_peers > GRPCOLL_MAX_PEERS) return -1;
  if (a->n_pieces == 0) return 0;
  GrpCollPullArgs k{};
  k.pieces = a->pieces;
  k.n_pieces = a->n_pieces;
  k.row_elems = a->row_elems;
  k.elem_size = a->elem_size;
  for (int i = 0; i < GRPCOLL_MAX_PEERS; ++i) {
    k.peer_ptrs[i] = a->peer_ptrs[i];
    k.peer_flags[i] = a->peer_flags[i];
  }
  k.wait_value = a->wait_value;
  k.n_peers = a->n_peers;
  k.dst = a->dst;
  k.reduce = a->reduce;
  const int grid = a->n_pieces < 64 ? a->n_pieces : 64;
  hipLaunchKernelGGL(grpcoll_pull_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)a->stream, k);
  return (int)hipGetLastError();
}

// ---- producer backpressure: wait until *flag >= value (acks) ----
__global__ void grpcoll_wait_kernel(const int* flag, int value) {
  if (threadIdx.x == 0) {
    while (__hip_atomic_load(flag, __ATOMIC_ACQUIRE,
                             __HIP_MEMORY_SCOPE_SYSTEM) < value) {
      __builtin_amdgcn_s_sleep(32);
    }
  }
}

extern "C" int magi_grpcoll_wait(const void* flag_ptr, int value,
                                 void* stream) {
  hipLaunchKernelGGL(grpcoll_wait_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const int*)flag_ptr, value);
  return (int)hipGetLastError();
}

// ---- consumer ack: atomicAdd(1) on each source peer's ack slot ----
struct GrpCollAckArgs {
  int* ack_ptrs[GRPCOLL_MAX_PEERS];
  int n;
};

__global__ void grpcoll_ack_kernel(GrpCollAckArgs a) {
  if (threadIdx.x == 0) {
    for (int i = 0; i < a.n; ++i) {
      if (a.ack_ptrs[i])
        __hip_atomic_fetch_add(a.ack_ptrs[i], 1, __ATOMIC_RELEASE,
                               __HIP_MEMORY_SCOPE_SYSTEM);
    }
  }
}

extern "C" int magi_grpcoll_ack(void* const* ack_ptrs, int n, void* stream) {
  if (n > GRPCOLL_MAX_PEERS) return -1;
  GrpCollAckArgs a{};
  for (int i = 0; i < n; ++i) a.ack_ptrs[i] = (int*)ack_ptrs[i];
  a.n = n;
  hipLaunchKernelGGL(grpcoll_ack_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, a);
  return (int)hipGetLastError();
}

// torch's caching allocator suballocates: an IPC handle names the BASE
// allocation, so peers need (handle, offset). Resolve the range start here.
extern "C" int magi_ipc_base(const void* dev_ptr, void** base,
                             unsigned long long* size) {
  hipDeviceptr_t b = 0;
  size_t sz = 0;
  hipError_t e = hipMemGetAddressRange(&b, &sz, (hipDeviceptr_t)dev_ptr);
  if (e != hipSuccess) return (int)e;
  *base = (void*)b;
  *size = (unsigned long long)sz;
  return 0;
}
