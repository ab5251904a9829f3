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

// ---- native range sort / reorder / unique (reference csrc/extensions/
// sort_and_reorder_ranges.cu:206, unique_consecutive_pairs.cu:153).
// Ranges are [N,2] int32 with N <= a few 10^4: one workgroup,
// LDS-resident bitonic sort on the packed (start,end) 64-bit key with the
// index in the low bits for stability; the host wrapper falls back to the
// torch ops above the LDS capacity (N > 8192).

#define SORT_CAP 8192

__global__ __launch_bounds__(1024) void sort_ranges_kernel(
    const int* ranges, int* out_idx, int n, int pow2) {
  __shared__ unsigned long long keys[SORT_CAP];
  // key = (start:25 | end:25 | idx:14)  -- starts/ends < 2^25 tokens,
  // idx < 2^14? N can reach 2^16: use (start,end) in the high 50 bits and
  // a 14-bit... safer: two LDS arrays {key64 = start<<32|end, idx}
  __shared__ int idxs[SORT_CAP];
  for (int i = threadIdx.x; i < pow2; i += blockDim.x) {
    if (i < n) {
      keys[i] = ((unsigned long long)(unsigned)ranges[2 * i] << 32) |
                (unsigned)ranges[2 * i + 1];
      idxs[i] = i;
    } else {
      keys[i] = ~0ull;
      idxs[i] = i;
    }
  }
  __syncthreads();
  for (int k = 2; k <= pow2; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      for (int i = threadIdx.x; i < pow2; i += blockDim.x) {
        const int ixj = i ^ j;
        if (ixj > i) {
          const bool up = (i & k) == 0;
          unsigned long long a = keys[i], b = keys[ixj];
          // stable: tie-break on original index
          bool swap = up ? (a > b || (a == b && idxs[i] > idxs[ixj]))
                         : (a < b || (a == b && idxs[i] < idxs[ixj]));
          if (swap) {
            keys[i] = b; keys[ixj] = a;
            int t = idxs[i]; idxs[i] = idxs[ixj]; idxs[ixj] = t;
          }
        }
      }
      __syncthreads();
    }
  }
  for (int i = threadIdx.x; i < n; i += blockDim.x) out_idx[i] = idxs[i];
}

extern "C" int magi_argsort_ranges(const void* ranges, void* out_idx, int n,
                                   void* stream) {
  if (n <= 0) return 0;
  if (n > SORT_CAP) return 2;  // caller falls back to the torch path
  int pow2 = 1;
  while (pow2 < n) pow2 <<= 1;
  hipLaunchKernelGGL(sort_ranges_kernel, dim3(1), dim3(1024), 0,
                     (hipStream_t)stream, (const int*)ranges, (int*)out_idx,
                     n, pow2);
  return (int)hipGetLastError();
}

__global__ __launch_bounds__(256) void reorder_ranges_kernel(
    const int* qr, const int* kr, const int* tm, const int* order,
    int* qro, int* kro, int* tmo, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const int s = order[i];
  qro[2 * i] = qr[2 * s];
  qro[2 * i + 1] = qr[2 * s + 1];
  kro[2 * i] = kr[2 * s];
  kro[2 * i + 1] = kr[2 * s + 1];
  tmo[i] = tm[s];
}

extern "C" int magi_reorder_ranges(const void* qr, const void* kr,
                                   const void* tm, const void* order,
                                   void* qro, void* kro, void* tmo, int n,
                                   void* stream) {
  if (n <= 0) return 0;
  hipLaunchKernelGGL(reorder_ranges_kernel, dim3((n + 255) / 256), dim3(256),
                     0, (hipStream_t)stream, (const int*)qr, (const int*)kr,
                     (const int*)tm, (const int*)order, (int*)qro, (int*)kro,
                     (int*)tmo, n);
  return (int)hipGetLastError();
}

// unique of SORTED [N,2] pairs: single-WG blocked scan over head flags
__global__ __launch_bounds__(1024) void unique_pairs_kernel(
    const int* ranges, int* uniq, int* inverse, int* count, int n) {
  __shared__ int pos[SORT_CAP];  // exclusive prefix of head flags
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    const bool head = (i == 0) || ranges[2 * i] != ranges[2 * i - 2] ||
                      ranges[2 * i + 1] != ranges[2 * i - 1];
    pos[i] = head ? 1 : 0;
  }
  __syncthreads();
  // simple per-thread serial scan by thread 0 (N <= 8192: ~8k adds, fine
  // for a once-per-plan op)
  if (threadIdx.x == 0) {
    int run = 0;
    for (int i = 0; i < n; ++i) {
      const int h = pos[i];
      pos[i] = run;
      run += h;
    }
    *count = run;
  }
  __syncthreads();
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    const bool head = (i == 0) || ranges[2 * i] != ranges[2 * i - 2] ||
                      ranges[2 * i + 1] != ranges[2 * i - 1];
    // inverse = inclusive prefix of head flags - 1
    const int u = pos[i] + (head ? 1 : 0) - 1;
    inverse[i] = u;
    if (head) {
      uniq[2 * u] = ranges[2 * i];
      uniq[2 * u + 1] = ranges[2 * i + 1];
    }
  }
}

extern "C" int magi_unique_pairs(const void* ranges, void* uniq,
                                 void* inverse, void* count, int n,
                                 void* stream) {
  if (n <= 0) return 0;
  if (n > SORT_CAP) return 2;
  hipLaunchKernelGGL(unique_pairs_kernel, dim3(1), dim3(1024), 0,
                     (hipStream_t)stream, (const int*)ranges, (int*)uniq,
                     (int*)inverse, (int*)count, n);
  return (int)hipGetLastError();
}
