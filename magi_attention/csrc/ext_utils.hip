// ext_utils.hip — magi_attn_ext native helpers (gfx950).
//
// KernelBarrier (reference csrc/extensions/kernel_barrier.cu:103): a device
// int32 counter; produce() increments it from the compute stream; a
// synchronize() spin kernel on the comm stream waits counter >= target —
// orders comm kernels after compute without a host sync
// (reference usage: functional/dist_attn.py:3054-3116).

#include <hip/hip_runtime.h>

#include "../../include/magi_ffa.h"

__global__ void barrier_produce_kernel(int* counter) {
  if (threadIdx.x == 0)
    __hip_atomic_fetch_add(counter, 1, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_AGENT);
}

__global__ void barrier_wait_kernel(const int* counter, int target) {
  if (threadIdx.x == 0) {
    while (__hip_atomic_load(counter, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT) < target)
      __builtin_amdgcn_s_sleep(8);
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
}

extern "C" int magi_kernel_barrier_produce(int32_t* counter, void* stream) {
  hipLaunchKernelGGL(barrier_produce_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, counter);
  return (int)hipGetLastError();
}

extern "C" int magi_kernel_barrier_synchronize(const int32_t* counter,
                                               int32_t target, void* stream) {
  hipLaunchKernelGGL(barrier_wait_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, counter, target);
  return (int)hipGetLastError();
}
