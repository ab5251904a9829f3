// range_ops.hip — HBM-bandwidth row gather / scatter-reduce kernels (gfx950).
//
// Replaces the reference's Triton range ops (common/range_op/_range_gather.py:126,
// _range_reduce.py:360, plus the fused correct_out_lse_kernel of
// functional/utils.py:371) with hand-written HIP. These are the pack/unpack
// engines of the group collectives: pure HBM-bound row copies/reductions over
// [T, h, d] tensors addressed by (range -> output start) tables.
//
// Roofline: HBM. Layout: rows are contiguous (h*d elements); a wave moves a
// row segment with 16-byte loads/stores per lane. Row lookup is a binary
// search over the per-range cumulative row offsets (n_ranges is small).

#include <hip/hip_runtime.h>

#include "../../include/magi_ffa.h"

#define DEV_INLINE __device__ __forceinline__

struct RangeParams {
  const char* input;
  char* output;
  const int* in_ranges;   // [n,2]
  const int* out_starts;  // [n]
  const float* in_lse;    // [rows_in, h] (lse reduce only)
  float* out_lse;         // [rows_out, h]
  long long n_ranges;
  long long row_bytes;    // bytes per row
  long long total_rows;
  int n_heads;
  int d;                  // elems per head (lse variant)
};

// cumulative row counts live in the first n+1 ints of a device scratch — to
// avoid extra allocations we recompute the range index by linear scan when n
// is tiny, else binary search over in_ranges lengths computed on the fly is
// impossible; instead each block handles ONE range (grid.y = range idx) and
// grid-strides over its rows. Ranges are typically few and row counts vary;
// blocks beyond a range's rows exit immediately.

template <typename VecT, int OP>  // OP: 0=copy, 1=sum(f32)
__global__ __launch_bounds__(256) void range_rows_kernel(RangeParams p) {
  const int ri = blockIdx.y;
  const int rs = p.in_ranges[2 * ri], re = p.in_ranges[2 * ri + 1];
  const int rows = re - rs;
  const int os = p.out_starts[ri];
  const long long vec_per_row = p.row_bytes / sizeof(VecT);
  // each block copies whole rows; row-major grid-stride
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const VecT* src = (const VecT*)(p.input + (size_t)(rs + row) * p.row_bytes);
    VecT* dst = (VecT*)(p.output + (size_t)(os + row) * p.row_bytes);
    for (long long i = threadIdx.x; i < vec_per_row; i += blockDim.x) {
      if (OP == 0) {
        dst[i] = src[i];
      } else {
        // f32 sum-reduce: VecT = float4. Ranges may target OVERLAPPING output
        // rows (several dst ranks returning partials for the same hosted
        // rows), and ranges run in concurrent blocks -> must be atomic
        // (-munsafe-fp-atomics => hardware global_atomic_add_f32).
        float4 a = *(const float4*)&src[i];
        float* d4 = (float*)&dst[i];
        atomicAdd(d4 + 0, a.x);
        atomicAdd(d4 + 1, a.y);
        atomicAdd(d4 + 2, a.z);
        atomicAdd(d4 + 3, a.w);
      }
    }
  }
}

// lse-weighted reduce (reference _range_reduce.py:239): merge partial
// (out,lse) rows into destination rows with online-softmax correction:
//   lse_m = max + log1p(exp(min-max)); out = w1*out1 + w2*out2.
// One wave per (row, head): d <= 128 fp32 elems handled by 64 lanes x ceil(d/64).
__global__ __launch_bounds__(256) void range_rows_lse_kernel(RangeParams p) {
  const int ri = blockIdx.y;
  const int rs = p.in_ranges[2 * ri], re = p.in_ranges[2 * ri + 1];
  const int rows = re - rs;
  const int os = p.out_starts[ri];
  const int h = p.n_heads, d = p.d;
  const int lane = threadIdx.x & 63;
  const int wavein = threadIdx.x >> 6;
  const int rowhead0 = (blockIdx.x * 4 + wavein);
  const long long n_rowheads = (long long)rows * h;
  for (long long rh = rowhead0; rh < n_rowheads; rh += (long long)gridDim.x * 4) {
    const int row = rh / h, hh = rh % h;
    const float* src = (const float*)p.input +
                       ((size_t)(rs + row) * h + hh) * d;
    float* dst = (float*)p.output + ((size_t)(os + row) * h + hh) * d;
    const float lse_in = p.in_lse[(size_t)(rs + row) * h + hh];
    float* lse_out_p = p.out_lse + (size_t)(os + row) * h + hh;
    const float lse_prev = *lse_out_p;
    const float mx = fmaxf(lse_in, lse_prev);
    const float mn = fminf(lse_in, lse_prev);
    const float lse_m = (mx == -INFINITY) ? -INFINITY : mx + log1pf(expf(mn - mx));
    const float w_new = (lse_in == -INFINITY) ? 0.f : expf(lse_in - lse_m);
    const float w_prev = (lse_prev == -INFINITY) ? 0.f : expf(lse_prev - lse_m);
    for (int i = lane; i < d; i += 64) {
      const float prev = (w_prev > 0.f) ? w_prev * dst[i] : 0.f;
      dst[i] = prev + w_new * src[i];
    }
    if (lane == 0) *lse_out_p = lse_m;
  }
}

static int launch_range(const magi_range_op_args* a, int op) {
  if (!a || !a->input || !a->output || !a->in_ranges || !a->out_starts)
    return -1;
  if (a->n_ranges <= 0) return 0;
  RangeParams p{};
  p.input = (const char*)a->input;
  p.output = (char*)a->output;
  p.in_ranges = a->in_ranges;
  p.out_starts = a->out_starts;
  p.in_lse = a->in_lse;
  p.out_lse = (float*)a->out_lse;
  p.n_ranges = a->n_ranges;
  p.row_bytes = a->row_elems * a->elem_size;
  p.total_rows = a->total_rows;
  p.n_heads = a->n_heads;
  p.d = a->n_heads > 0 ? (int)(a->row_elems / a->n_heads) : 0;
  hipStream_t s = (hipStream_t)a->stream;
  // rows per range unknown on host: size grid.x for the average
  long long avg_rows = (a->total_rows + a->n_ranges - 1) / a->n_ranges;
  unsigned gx = (unsigned)min((long long)2048, max((long long)1, avg_rows));
  dim3 grid(gx, (unsigned)a->n_ranges), block(256);
  if (op == 2) {
    if (!a->in_lse || !a->out_lse) return -3;
    hipLaunchKernelGGL(range_rows_lse_kernel, grid, block, 0, s, p);
  } else if (op == 1) {
    if (p.row_bytes % 16 != 0) return -2;
    hipLaunchKernelGGL((range_rows_kernel<float4, 1>), grid, block, 0, s, p);
  } else {
    if (p.row_bytes % 16 == 0)
      hipLaunchKernelGGL((range_rows_kernel<float4, 0>), grid, block, 0, s, p);
    else if (p.row_bytes % 4 == 0)
      hipLaunchKernelGGL((range_rows_kernel<float, 0>), grid, block, 0, s, p);
    else
      hipLaunchKernelGGL((range_rows_kernel<char, 0>), grid, block, 0, s, p);
  }
  return (int)hipGetLastError();
}

extern "C" int magi_range_gather(const magi_range_op_args* a) {
  return launch_range(a, 0);
}

extern "C" int magi_range_reduce(const magi_range_op_args* a) {
  return launch_range(a, a->reduce_op);
}

// ------------------------------------------------------------------
// fused correct_out_lse (reference functional/utils.py:371): whole-tensor
// merge of two partial (out,lse) sets, out1/lse1 updated in place.
// ------------------------------------------------------------------
__global__ __launch_bounds__(256) void correct_out_lse_kernel(
    float* out1, float* lse1, const float* out2, const float* lse2,
    long long n_rowheads, int d) {
  const int lane = threadIdx.x & 63;
  const int wavein = threadIdx.x >> 6;
  for (long long rh = blockIdx.x * 4 + wavein; rh < n_rowheads;
       rh += (long long)gridDim.x * 4) {
    const float l1 = lse1[rh], l2 = lse2[rh];
    const float mx = fmaxf(l1, l2), mn = fminf(l1, l2);
    const float lm = (mx == -INFINITY) ? -INFINITY : mx + log1pf(expf(mn - mx));
    const float w1 = (l1 == -INFINITY) ? 0.f : expf(l1 - lm);
    const float w2 = (l2 == -INFINITY) ? 0.f : expf(l2 - lm);
    float* o1 = out1 + rh * d;
    const float* o2 = out2 + rh * d;
    for (int i = lane; i < d; i += 64) o1[i] = w1 * o1[i] + w2 * o2[i];
    if (lane == 0) lse1[rh] = lm;
  }
}

extern "C" int magi_correct_out_lse(const magi_correct_args* a) {
  if (!a || !a->out1 || !a->lse1 || !a->out2 || !a->lse2) return -1;
  const long long n = a->total_rows * a->n_heads;
  if (n == 0) return 0;
  unsigned gx = (unsigned)min((long long)4096, (n + 3) / 4);
  hipLaunchKernelGGL(correct_out_lse_kernel, dim3(gx), dim3(256),
                     0, (hipStream_t)a->stream, (float*)a->out1, a->lse1,
                     (const float*)a->out2, a->lse2, n, a->d);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------------
// attention-sink postprocess + dsink (reference
// flash_fwd_postprocess_kernel.h:39, flash_bwd_preprocess_kernel.h dsink
// path): see include/magi_ffa.h for the math.
// ------------------------------------------------------------------
template <bool OUT_F32, bool SSH>
__global__ __launch_bounds__(256) void sink_postprocess_kernel(
    magi_sink_args a) {
  const int lane = threadIdx.x & 63;
  const int wavein = threadIdx.x >> 6;
  const long long n_rowheads = a.total_rows * a.n_heads;
  for (long long rh = blockIdx.x * 4 + wavein; rh < n_rowheads;
       rh += (long long)gridDim.x * 4) {
    const long long row = rh / a.n_heads;
    const int h = rh % a.n_heads;
    // lse_sink = logsumexp over the (<=8) sink logits of this (row, head)
    float mx = -INFINITY;
    float sv[8];
    for (int j = 0; j < a.s_sink; ++j) {
      sv[j] = SSH ? a.sink[(row * a.s_sink + j) * a.n_heads + h]
                  : a.sink[(long long)j * a.n_heads + h];
      mx = fmaxf(mx, sv[j]);
    }
    float l = 0.f;
    for (int j = 0; j < a.s_sink; ++j) l += expf(sv[j] - mx);
    const float lse_sink = (mx == -INFINITY) ? -INFINITY : mx + logf(l);

    float* lse_p = a.lse + rh;
    const float lse_old = *lse_p;
    float w, lse_new;
    if (lse_old == -INFINITY) {
      w = 0.f;
      lse_new = lse_sink;
    } else if (lse_sink == -INFINITY) {
      w = 1.f;
      lse_new = lse_old;
    } else {
      const float m = fmaxf(lse_old, lse_sink);
      lse_new = m + logf(expf(lse_old - m) + expf(lse_sink - m));
      w = expf(lse_old - lse_new);
    }
    if (lane == 0) *lse_p = lse_new;
    if (w != 1.f) {
      if (OUT_F32) {
        float* o = (float*)a.out + rh * a.d;
        for (int i = lane; i < a.d; i += 64) o[i] *= w;
      } else {
        __bf16* o = (__bf16*)a.out + rh * a.d;
        for (int i = lane; i < a.d; i += 64)
          o[i] = (__bf16)((float)o[i] * w);
      }
    }
  }
}

extern "C" int magi_ffa_sink_postprocess(const magi_sink_args* a) {
  if (!a || !a->out || !a->lse || !a->sink) return -1;
  if (a->s_sink <= 0 || a->s_sink > 8) return -2;
  const long long n = a->total_rows * a->n_heads;
  if (n == 0) return 0;
  unsigned gx = (unsigned)min((long long)4096, (n + 3) / 4);
  hipStream_t s = (hipStream_t)a->stream;
  if (a->out_is_fp32) {
    if (a->ssh)
      hipLaunchKernelGGL((sink_postprocess_kernel<true, true>), dim3(gx),
                         dim3(256), 0, s, *a);
    else
      hipLaunchKernelGGL((sink_postprocess_kernel<true, false>), dim3(gx),
                         dim3(256), 0, s, *a);
  } else {
    if (a->ssh)
      hipLaunchKernelGGL((sink_postprocess_kernel<false, true>), dim3(gx),
                         dim3(256), 0, s, *a);
    else
      hipLaunchKernelGGL((sink_postprocess_kernel<false, false>), dim3(gx),
                         dim3(256), 0, s, *a);
  }
  return (int)hipGetLastError();
}

// "sh": one block per (row-block, head); block-partial sums in LDS, one
// atomicAdd per (block, j) — contention is blocks*h*s_sink, not rows.
__global__ __launch_bounds__(256) void dsink_sh_kernel(magi_sink_args a) {
  __shared__ float part[8][64];
  const int h = blockIdx.y;
  const int tid = threadIdx.x;
  const long long row0 = (long long)blockIdx.x * 256;
  float p[8];
  for (int j = 0; j < a.s_sink; ++j) p[j] = 0.f;
  const long long row = row0 + tid;
  if (row < a.total_rows) {
    const float lse = a.lse[row * a.n_heads + h];
    const float dps = a.dpsum[row * a.n_heads + h];
    if (lse != INFINITY && lse != -INFINITY) {
      for (int j = 0; j < a.s_sink; ++j)
        p[j] = -expf(a.sink[(long long)j * a.n_heads + h] - lse) * dps;
    }
  }
  // wave reduce then cross-wave via LDS
  const int lane = tid & 63;
  const int w = tid >> 6;
  for (int j = 0; j < a.s_sink; ++j) {
    float x = p[j];
    for (int off = 32; off; off >>= 1)
      x += __shfl_down(x, off, 64);
    if (lane == 0) part[j][w] = x;
  }
  __syncthreads();
  if (tid < a.s_sink) {
    float x = part[tid][0] + part[tid][1] + part[tid][2] + part[tid][3];
    atomicAdd(a.dsink + (long long)tid * a.n_heads + h, x);
  }
}

__global__ __launch_bounds__(256) void dsink_ssh_kernel(magi_sink_args a) {
  const long long n = a.total_rows * a.n_heads;
  for (long long rh = (long long)blockIdx.x * 256 + threadIdx.x; rh < n;
       rh += (long long)gridDim.x * 256) {
    const long long row = rh / a.n_heads;
    const int h = rh % a.n_heads;
    const float lse = a.lse[rh];
    const float dps = a.dpsum[rh];
    const bool live = lse != INFINITY && lse != -INFINITY;
    for (int j = 0; j < a.s_sink; ++j) {
      const long long idx = (row * a.s_sink + j) * a.n_heads + h;
      a.dsink[idx] = live ? -expf(a.sink[idx] - lse) * dps : 0.f;
    }
  }
}

extern "C" int magi_ffa_dsink(const magi_sink_args* a) {
  if (!a || !a->lse || !a->sink || !a->dsink || !a->dpsum) return -1;
  if (a->s_sink <= 0 || a->s_sink > 8) return -2;
  if (a->total_rows == 0) return 0;
  hipStream_t s = (hipStream_t)a->stream;
  if (a->ssh) {
    const long long n = a->total_rows * a->n_heads;
    unsigned gx = (unsigned)min((long long)4096, (n + 255) / 256);
    hipLaunchKernelGGL(dsink_ssh_kernel, dim3(gx), dim3(256), 0, s, *a);
  } else {
    unsigned gx = (unsigned)((a->total_rows + 255) / 256);
    hipLaunchKernelGGL(dsink_sh_kernel, dim3(gx, a->n_heads), dim3(256), 0, s,
                       *a);
  }
  return (int)hipGetLastError();
}
