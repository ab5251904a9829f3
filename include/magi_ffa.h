/* magi_ffa.h — C-ABI boundary of the MI355X-native flex-flash-attention engine.
 *
 * These entry points are the drop-in replacement for the reference's JIT'd FFA
 * kernel module surface (reference: magi_attention/functional/flex_flash_attn.py:261-290
 * `mod.fwd`, :527-554 `mod.bwd`, resolved by _flex_flash_attn_jit.py:456
 * `get_ffa_jit_mod`). The host side (magi_attention/functional/flex_flash_attn.py
 * in THIS repo) binds them via ctypes; INTEGRATION.md shows the binding a
 * maintainer would add.
 *
 * Conventions: all device pointers; q/k/v bf16 row-major [tokens, heads, dim];
 * ranges int32 [n,2] half-open; attn_type_map int32 (0=full, 1=causal,
 * 2=inv_causal, 3=bi_causal; semantics = reference flex_flash_attn.py:1247-1341);
 * stream is a hipStream_t. All functions return 0 on success, nonzero hipError_t
 * or negative validation error otherwise.
 */
#ifndef MAGI_FFA_H
#define MAGI_FFA_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct magi_ffa_fwd_args {
  const void* q;            /* bf16 [total_q, hq, d] */
  const void* k;            /* bf16 [total_k, hk, d] */
  const void* v;            /* bf16 [total_k, hk, d] */
  void* out;                /* f32 (accumulate/merge mode) or bf16 [total_q, hq, d] */
  float* lse;               /* f32 [total_q, hq], caller-initialised to -inf */
  const int32_t* q_ranges;  /* [n, 2] */
  const int32_t* k_ranges;  /* [n, 2] */
  const int32_t* attn_type_map; /* [n] or NULL (= all full) */
  int32_t* locks;           /* zeroed int32 [ceil(total_q/128) * hq]; NULL iff
                               disable_atomic_reduction */
  float* max_logits;        /* f32 [hq] init -inf, atomic-max of the scaled
                               (softcapped) logits per head; NULL = off
                               (reference return_max_logits) */
  const int32_t* qk_starts; /* auto_range_merge: [n_ranges+1]; q_ranges are
                               UNIQUE, k segments per unique q range; NULL =
                               one k range per q range */
  int64_t n_ranges;
  int64_t total_q;
  int64_t total_k;
  int32_t hq;
  int32_t hk;
  int32_t d;                /* head dim: 64 or 128 */
  int32_t max_seqlen_q;     /* upper bound on max q-range length */
  float softmax_scale;
  float softcap;            /* 0 = disabled; else score=softcap*tanh(s*scale/softcap),
                               softmax scale becomes softcap (reference
                               mainloop_fwd...hpp:466-489) */
  int32_t out_is_fp32;      /* 1: out is f32 accumulator (merge into it);
                               0: bf16 direct store (requires disable_atomic_reduction) */
  int32_t disable_atomic_reduction; /* 1: q_ranges disjoint, skip lock/merge */
  int32_t cu_margin;        /* CUs left free for comm kernels (sm_margin analogue) */
  void* stream;             /* hipStream_t */
} magi_ffa_fwd_args;

int magi_ffa_fwd(const magi_ffa_fwd_args* args);

typedef struct magi_ffa_bwd_args {
  const void* dout;         /* bf16 [total_q, hq, d] */
  const void* q;            /* bf16 [total_q, hq, d] */
  const void* k;            /* bf16 [total_k, hk, d] */
  const void* v;            /* bf16 [total_k, hk, d] */
  const void* out;          /* bf16 or f32 [total_q, hq, d] (forward output) */
  const float* lse;         /* f32 [total_q, hq] */
  float* dq;                /* f32 [total_q, hq, d], zeroed, atomic-accumulated */
  float* dk;                /* f32 [total_k, hk, d], zeroed, atomic-accumulated */
  float* dv;                /* f32 [total_k, hk, d], zeroed, atomic-accumulated */
  float* dpsum;             /* f32 [total_q, hq] workspace = rowsum(dO*O) */
  const int32_t* q_ranges;  /* [n, 2] */
  const int32_t* k_ranges;  /* [n, 2] */
  const int32_t* attn_type_map; /* [n] or NULL */
  const int32_t* seg_starts;    /* auto_range_merge: [n_ranges+1] segment map
                                   (dq: q->k segments; dkv: k->q segments);
                                   NULL = one segment per range */
  int64_t n_ranges;
  int64_t total_q;
  int64_t total_k;
  int32_t hq;
  int32_t hk;
  int32_t d;
  int32_t max_seqlen_k;     /* upper bound on max k-range length */
  int32_t out_is_fp32;      /* dtype of `out` above */
  float softmax_scale;
  float softcap;
  int32_t cu_margin;
  void* stream;
} magi_ffa_bwd_args;

/* dPsum preprocess: dpsum[t,h] = sum_d dout[t,h,d] * out[t,h,d]
 * (reference flash_bwd_preprocess_kernel.h:42). */
int magi_ffa_bwd_preprocess(const magi_ffa_bwd_args* args);

/* 5-matmul backward (reference flash_bwd_kernel_sm90.h:41), split into two
 * INDEPENDENT passes so the host may run them on separate streams
 * concurrently: the dq pass (q-outer, dQ register-accumulated) and the dkv
 * pass (k-outer, dK/dV register-accumulated). Both require dpsum filled.
 * magi_ffa_bwd runs both sequentially on args->stream. */
int magi_ffa_bwd_dq(const magi_ffa_bwd_args* args);
int magi_ffa_bwd_dkv(const magi_ffa_bwd_args* args);
/* dv/dk pass split of magi_ffa_bwd_dkv: each half re-derives S but fits
 * 2 waves/SIMD on gfx950; launching both is result-identical to the fused
 * kernel up to fp32 atomic-add ordering. */
int magi_ffa_bwd_dv(const magi_ffa_bwd_args* args);
int magi_ffa_bwd_dk(const magi_ffa_bwd_args* args);
int magi_ffa_bwd(const magi_ffa_bwd_args* args);

/* ------------------------------------------------------------------ *
 * Range row ops (reference common/range_op/_range_*.py — Triton there,
 * hand-written HIP here; pack/unpack engine of the group collectives). *
 * ------------------------------------------------------------------ */

typedef struct magi_range_op_args {
  const void* input;        /* [T_in, row_elems] */
  void* output;             /* [T_out, row_elems] */
  const int32_t* in_ranges;   /* [n,2] rows of input  */
  const int32_t* out_starts;  /* [n]   start row in output for each range */
  const float* in_lse;      /* optional [T_in, h] (lse-weighted reduce) */
  float* out_lse;           /* optional [T_out, h] */
  int64_t n_ranges;
  int64_t row_elems;        /* elements per row (h*d) */
  int64_t total_rows;       /* total rows moved (sum of range lengths) */
  int32_t elem_size;        /* bytes per element: 2 (bf16) or 4 (f32) */
  int32_t n_heads;          /* for lse variant: row_elems = n_heads*d */
  int32_t reduce_op;        /* 0=copy(gather/scatter), 1=sum, 2=lse-weighted */
  void* stream;
} magi_range_op_args;

/* output[out_starts[i] + j] = input[in_ranges[i].start + j] (copy), or
 * output rows += / lse-merge input rows (reduce). */
int magi_range_gather(const magi_range_op_args* args);
int magi_range_reduce(const magi_range_op_args* args);

/* Attention-sink postprocess (reference flash_fwd_postprocess_kernel.h:39
 * FlashAttnFwdPostprocess / flash_bwd_preprocess_kernel.h dsink path):
 * value-less learnable logits folded into (out, lse) AFTER the (possibly
 * multi-slice, atomic-merged) forward, so they enter the softmax denominator
 * exactly once. Layouts: ssh=0 -> sink[s_sink, h]; ssh=1 -> sink[T, s_sink, h].
 */
typedef struct magi_sink_args {
  void* out;            /* [T, h, d] bf16 (out_is_fp32=0) or f32, in/out */
  float* lse;           /* [T, h], in/out (fwd) / in (bwd) */
  const float* sink;    /* sink logits, f32 */
  float* dsink;         /* same shape as sink, f32, zero-initialised (bwd) */
  const float* dpsum;   /* [T, h] (bwd) */
  int64_t total_rows;   /* T */
  int32_t n_heads;
  int32_t d;
  int32_t s_sink;       /* <= 8 */
  int32_t ssh;          /* 0 = "sh", 1 = "ssh" */
  int32_t out_is_fp32;
  void* stream;
} magi_sink_args;

/* fwd: lse_sink = log(sum_j exp(sink_j)); live rows: out *= exp(lse-lse'),
 * lse' = log(exp(lse)+exp(lse_sink)); empty rows (lse=-inf): out=0, lse=lse_sink. */
int magi_ffa_sink_postprocess(const magi_sink_args* args);
/* bwd: dsink_j (+)= -exp(sink_j - lse_row) * dpsum_row (summed over rows for
 * "sh", per row for "ssh"). */
int magi_ffa_dsink(const magi_sink_args* args);

/* Fused merge of two partial (out, lse) sets (reference functional/utils.py:371
 * correct_out_lse_kernel): out1/lse1 updated in place with out2/lse2 merged in. */
typedef struct magi_correct_args {
  void* out1;               /* f32 [T, h, d], in/out */
  float* lse1;              /* f32 [T, h], in/out */
  const void* out2;         /* f32 [T, h, d] */
  const float* lse2;        /* f32 [T, h] */
  int64_t total_rows;       /* T */
  int32_t n_heads;
  int32_t d;
  void* stream;
} magi_correct_args;

int magi_correct_out_lse(const magi_correct_args* args);

/* ------------------------------------------------------------------ *
 * magi_attn_ext surface helpers (reference csrc/extensions/)           *
 * ------------------------------------------------------------------ */

/* GPU spin barrier (reference extensions/kernel_barrier.cu:103): counter in
 * device memory; produce() increments from one stream, synchronize() launches
 * a 1-thread spin kernel on another stream that waits counter >= target. */
int magi_kernel_barrier_produce(int32_t* counter, void* stream);
int magi_kernel_barrier_synchronize(const int32_t* counter, int32_t target,
                                    void* stream);

/* native range utilities (reference csrc/extensions/
   sort_and_reorder_ranges.cu, unique_consecutive_pairs.cu): single-WG HIP
   kernels for the planner regime (N <= 8192); rc=2 => caller falls back */
int magi_argsort_ranges(const void* ranges, void* out_idx, int n,
                        void* stream);
int magi_reorder_ranges(const void* q_ranges, const void* k_ranges,
                        const void* attn_type_map, const void* order,
                        void* q_out, void* k_out, void* t_out, int n,
                        void* stream);
int magi_unique_pairs(const void* sorted_ranges, void* uniq, void* inverse,
                      void* count, int n, void* stream);

/* native grpcoll transport (HIP-IPC pull over xGMI; csrc/grpcoll.hip) */
int magi_ipc_get_handle(const void* dev_ptr, void* out_handle64);
int magi_ipc_open(const void* handle64, void** out_ptr);
int magi_ipc_close(void* ptr);
int magi_ipc_base(const void* dev_ptr, void** base, unsigned long long* size);
int magi_grpcoll_signal(void* flag_ptr, int value, void* stream);
int magi_grpcoll_wait(const void* flag_ptr, int value, void* stream);
int magi_grpcoll_ack(void* const* ack_ptrs, int n, void* stream);
struct magi_grpcoll_pull_args;
int magi_grpcoll_pull(const struct magi_grpcoll_pull_args* args);

/* Index-attention (token-gather) forward — the reference's
 * index_attn_indices direct-to-kernel path (flex_flash_attn.py:1358-1390).
 * indices_2d[i] lists the GLOBAL K rows attended by q token row i, shared
 * by all hq query heads of that row; -1 = contiguous tail padding. KV heads
 * are folded into the K row dimension (hk must be 1). Forward only. */
typedef struct magi_ffa_index_args {
  const void* q;              /* bf16 [total_q, hq, d] */
  const void* k;              /* bf16 [total_k, 1, d] */
  const void* v;              /* bf16 [total_k, 1, d] */
  void* out;                  /* bf16 or f32 [total_q, hq, d], zero-init */
  float* lse;                 /* f32 [total_q, hq], caller-initialised -inf */
  const int32_t* indices_2d;  /* [total_q, max_topk] global K row ids, -1 pad */
  int64_t total_q;
  int64_t total_k;
  int32_t max_topk;           /* multiple of 64 */
  int32_t hq;
  int32_t hk;                 /* must be 1 */
  int32_t d;                  /* 64 or 128 */
  float softmax_scale;
  float softcap;
  int32_t out_is_fp32;
  void* stream;               /* hipStream_t */
} magi_ffa_index_args;

int magi_ffa_fwd_index(const magi_ffa_index_args* args);

/* Version/identity probe so tests can verify the native library is loaded. */
int magi_ffa_abi_version(void);

#ifdef __cplusplus
} /* extern "C" */
#endif

#endif /* MAGI_FFA_H */
