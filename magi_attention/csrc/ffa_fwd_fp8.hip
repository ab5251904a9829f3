// ffa_fwd_fp8.hip — fp8 (OCP e4m3) flex-flash-attention FORWARD (gfx950).
//
// MI355X-native EXTENSION: the reference has no fp8 compute path
// (MagiAttentionPrecision = bf16/fp16/fp32/fp64, common/enum.py:149); this
// implements BASELINE.json config 5 on the CDNA4 fp8 MFMAs
// (v_mfma_f32_32x32x16_fp8_fp8, fp32 accumulate — "bf16 accumulate" in the
// config wording is subsumed by the wider fp32 accumulator).
//
// Numerics follow the reference's fp8 softmax convention (softmax.h:151-153,
// 326-331): subtract max_offset = 8 in the exp2 domain so P lands in
// [0, 256] (inside e4m3 range, max 448), and add it back into the lse.
// Parity is pinned against the bf16/fp64 oracle with fp8-calibrated
// thresholds (tests/test_ffa_fp8_gpu.py) — documented as an extension, not
// parity-at-1e-3 (SURVEY.md §8c fp8 caveat).
//
// Structure: the forward v1 shape — 4 waves x 32 q rows, KV tiles of 32
// staged cooperatively (K rows 16B-swizzled for ds_read_b64 A-frags; V into a
// byte-transposed tile), double-buffered register pipeline. P is packed to
// fp8 in-register (ONE permlane32_swap per 16-k tile).

#include <hip/hip_runtime.h>
#include <math.h>

#include "../../include/magi_ffa.h"

#define FP8_BM 128
#define FP8_BN 32
#define LOCK_GRAN 128

using f32x16 = __attribute__((ext_vector_type(16))) float;
using u8 = unsigned char;

#define DEV_INLINE __device__ __forceinline__

DEV_INLINE float fast_exp2(float x) { return __builtin_amdgcn_exp2f(x); }

DEV_INLINE float warp_xor32(float v) { return __shfl_xor(v, 32, 64); }
DEV_INLINE int crow(int r, int hi) { return (r & 3) + 8 * (r >> 2) + 4 * hi; }

DEV_INLINE u8 to_fp8(float x) {
  // f32 -> e4m3 via the packed convert (lower byte)
  union {
    short s;
    u8 b[2];
  } r;
  r.s = __builtin_amdgcn_cvt_pk_fp8_f32(x, 0.f, 0, false);
  return r.b[0];
}

struct Fp8FwdParams {
  const u8* q;
  const u8* k;
  const u8* v;
  float* out_f32;
  float* lse;
  const int* q_ranges;
  const int* k_ranges;
  const int* attn_type_map;
  int* locks;
  float* max_logits;
  int hq, hk, gqa;
  float scale;
  long long total_q, total_k;
};

template <int D>
__global__ __launch_bounds__(256, 2) void ffa_fwd_fp8_kernel(Fp8FwdParams p) {
  constexpr int DF = D / 16;   // 16-wide d fragments per MFMA chain
  constexpr int DT = D / 32;
  constexpr int ROWB = D;      // bytes per row (1 B per element)
  const int ri = blockIdx.y;
  const int h = blockIdx.z;
  const int qs = p.q_ranges[2 * ri], qe = p.q_ranges[2 * ri + 1];
  const int m0 = qs + blockIdx.x * FP8_BM;
  if (m0 >= qe) return;
  const int ks = p.k_ranges[2 * ri], ke = p.k_ranges[2 * ri + 1];
  const int atype = p.attn_type_map ? p.attn_type_map[ri] : 0;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo32 = lane & 31;
  const int hi = lane >> 5;

  const int q0 = m0 + wave * 32;
  const int qrow = q0 + lo32;
  const bool qvalid = (qrow < qe) && (q0 < qe);
  const bool qvalid_any = q0 < qe;
  const int qclamp = qvalid ? qrow : (qe - 1);

  const float sl2 = p.scale * 1.4426950408889634f;
  constexpr float MAX_OFFSET = 8.f;  // reference softmax.h:151 fp8 offset

  int n_lo = ks, n_hi = ke;
  if (ke > ks && q0 < qe) {
    const int qhiw = min(q0 + 31, qe - 1);
    if (atype == 1 || atype == 3) n_hi = min(n_hi, qhiw + (ke - qe) + 1);
    if (atype == 2 || atype == 3) n_lo = max(n_lo, q0 + (ks - qs));
  } else {
    n_hi = n_lo;
  }
  int b_lo = ks, b_hi = ke;
  if (ke > ks && m0 < qe) {
    const int qhib = min(m0 + FP8_BM - 1, qe - 1);
    if (atype == 1 || atype == 3) b_hi = min(b_hi, qhib + (ke - qe) + 1);
    if (atype == 2 || atype == 3) b_lo = max(b_lo, m0 + (ks - qs));
  } else {
    b_hi = b_lo;
  }

  // Q fragments: 8 fp8 bytes per 16-d slice
  long qf[DF];
  {
    const u8* qp = p.q + (size_t)qclamp * p.hq * D + (size_t)h * D;
#pragma unroll
    for (int dd = 0; dd < DF; ++dd)
      qf[dd] = *(const long*)(qp + dd * 16 + hi * 8);
  }

  const int kh = h / p.gqa;
  const size_t k_pitch = (size_t)p.hk * D;
  const u8* kbase = p.k + (size_t)kh * D;
  const u8* vbase = p.v + (size_t)kh * D;

  float m_run = -INFINITY;
  float l_run = 0.f;
  f32x16 acc_o[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) acc_o[dt] = (f32x16)(0.f);

  // LDS: K rows (8-B swizzled) + byte-transposed V, double-buffered
  __shared__ __attribute__((aligned(16))) char smem8[2 * FP8_BN * D +
                                                     2 * D * 40];
  auto lds_k = [&](int buf) -> u8* { return (u8*)(smem8 + buf * FP8_BN * D); };
  auto lds_vt = [&](int buf) -> u8(*)[40] {
    return (u8(*)[40])(smem8 + 2 * FP8_BN * D + buf * D * 40);
  };
  // 8-B-granular swizzle keeps ds_read_b64 A-frag groups spread
  auto swz = [](int row, int byte_off) {
    return byte_off ^ ((row & 7) << 3);
  };

  constexpr int CPR = D / 16;     // 16-B chunks per row
  constexpr int RPP = 256 / CPR;  // rows per pass (32 for D=128)
  constexpr int NPASS = (FP8_BN + RPP - 1) / RPP;
  const int srow = threadIdx.x / CPR;
  const int scol = threadIdx.x % CPR;
  using u8x16 = __attribute__((ext_vector_type(16))) u8;
  // two register staging sets (see ffa_fwd.hip): loads for tile t+2 fly
  // under tile t's compute, write_stage at loop TOP waits on loads issued a
  // full iteration earlier
  u8x16 kregA[NPASS], vregA[NPASS], kregB[NPASS], vregB[NPASS];

  auto issue_loads = [&](int n0, auto& kreg, auto& vreg) {
#pragma unroll
    for (int pass = 0; pass < NPASS; ++pass) {
      const int r = pass * RPP + srow;
      const int kr = min(n0 + min(r, FP8_BN - 1), ke - 1);
      kreg[pass] = *(const u8x16*)(kbase + (size_t)kr * k_pitch + scol * 16);
      vreg[pass] = *(const u8x16*)(vbase + (size_t)kr * k_pitch + scol * 16);
    }
  };
  auto write_stage = [&](int buf, auto& kreg, auto& vreg) {
#pragma unroll
    for (int pass = 0; pass < NPASS; ++pass) {
      const int r = pass * RPP + srow;
      if (r < FP8_BN) {
        // K row image: two swizzled 8-B halves of the 16-B chunk
        *(long*)((char*)lds_k(buf) + swz(r, r * ROWB + scol * 16)) =
            ((const long*)&kreg[pass])[0];
        *(long*)((char*)lds_k(buf) + swz(r, r * ROWB + scol * 16 + 8)) =
            ((const long*)&kreg[pass])[1];
        const int bs = (scol & 3) << 3;  // bank-spread on the k index
#pragma unroll
        for (int e = 0; e < 16; ++e)
          lds_vt(buf)[scol * 16 + e][r ^ bs] = vreg[pass][e];
      }
    }
  };

  int cur = 0;
  if (b_lo < b_hi) {
    issue_loads(b_lo, kregA, vregA);
    write_stage(0, kregA, vregA);
    issue_loads(b_lo + FP8_BN, kregA, vregA);
  }
  __syncthreads();

  auto iter_body = [&](int n0, auto& kreg_w, auto& vreg_w, auto& kreg_l,
                       auto& vreg_l) {
    const bool has_next = n0 + FP8_BN < b_hi;
    if (has_next) write_stage(cur ^ 1, kreg_w, vreg_w);
    issue_loads(n0 + 2 * FP8_BN, kreg_l, vreg_l);  // clamped overrun is safe
    const bool live = (n0 + FP8_BN > n_lo) && (n0 < n_hi) && qvalid_any;
    if (!live) {
      __syncthreads();
      cur ^= 1;
      return;
    }

    // ---- S^T = K Q^T (swapped; A = K rows from LDS) ----
    f32x16 s = (f32x16)(0.f);
#pragma unroll
    for (int dd = 0; dd < DF; ++dd) {
      long kf = *(const long*)((const char*)lds_k(cur) +
                               swz(lo32, lo32 * ROWB + dd * 16 + hi * 8));
      s = __builtin_amdgcn_mfma_f32_32x32x16_fp8_fp8(kf, qf[dd], s, 0, 0, 0);
    }

    const bool interior =
        (q0 + 31 < qe) && (n0 >= n_lo) && (n0 + FP8_BN <= n_hi) &&
        !((atype == 1 || atype == 3) && (n0 + FP8_BN - 1 > q0 + (ke - qe))) &&
        !((atype == 2 || atype == 3) && (n0 < q0 + 31 + (ks - qs)));
    float t[16];
    float mx = -INFINITY;
    if (interior) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        t[r] = s[r] * sl2;
        mx = fmaxf(mx, t[r]);
      }
    } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kk = n0 + crow(r, hi);
        bool ok = qvalid && (kk >= n_lo) && (kk < n_hi);
        if (atype == 1 || atype == 3) ok = ok && (kk - qrow <= ke - qe);
        if (atype == 2 || atype == 3) ok = ok && (kk - qrow >= ks - qs);
        t[r] = ok ? s[r] * sl2 : -INFINITY;
        mx = fmaxf(mx, t[r]);
      }
    }
    mx = fmaxf(mx, warp_xor32(mx));

    const float m_new = fmaxf(m_run, mx);
    const float m_use = (m_new == -INFINITY) ? 0.f : m_new;
    const float alpha = (m_run == -INFINITY) ? 0.f : fast_exp2(m_run - m_use);
    m_run = m_new;

    float pr[16];
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      // fp8 convention: exp2 output in [0, 256] (max_offset = 8)
      pr[r] = fast_exp2(t[r] - m_use + MAX_OFFSET);
      psum += pr[r];
    }
    l_run = l_run * alpha + (psum + warp_xor32(psum));

    if (__any(alpha != 1.f)) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int src = crow(r, hi);
        const float aq = __uint_as_float(
            __builtin_amdgcn_ds_bpermute(src << 2, __float_as_uint(alpha)));
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) acc_o[dt][r] *= aq;
      }
    }

    // ---- P -> fp8 A-fragments (pack 4 + ONE permlane swap per 16-k) ----
    long pa[2];
#pragma unroll
    for (int tt = 0; tt < 2; ++tt) {
      union {
        u8 b[4];
        unsigned u;
      } A, B;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        A.b[j] = to_fp8(pr[8 * tt + j]);
        B.b[j] = to_fp8(pr[8 * tt + 4 + j]);
      }
      auto r2 = __builtin_amdgcn_permlane32_swap(A.u, B.u, false, false);
      pa[tt] = ((long)(unsigned)r2[1] << 32) | (unsigned)r2[0];
    }

    // ---- PV from the byte-transposed V tile ----
#pragma unroll
    for (int tt = 0; tt < 2; ++tt) {
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        // bank-spread XOR must match the write side: bs(d) = ((d>>4)&3)<<3
        const int kbs = (((dt * 32 + lo32) >> 4) & 3) << 3;
        long bv = *(const long*)(
            &lds_vt(cur)[dt * 32 + lo32][(16 * tt + 8 * hi) ^ kbs]);
        acc_o[dt] =
            __builtin_amdgcn_mfma_f32_32x32x16_fp8_fp8(pa[tt], bv, acc_o[dt], 0, 0, 0);
      }
    }
    __syncthreads();
    cur ^= 1;
  };

  {
    int n0 = b_lo;
    while (n0 < b_hi) {
      iter_body(n0, kregA, vregA, kregB, vregB);
      n0 += FP8_BN;
      if (n0 >= b_hi) break;
      iter_body(n0, kregB, vregB, kregA, vregA);
      n0 += FP8_BN;
    }
  }

  if (p.max_logits) {
    float ml = m_run * 0.6931471805599453f;  // see bf16 fwd epilogue
#pragma unroll
    for (int off = 32; off; off >>= 1)
      ml = fmaxf(ml, __shfl_xor(ml, off, 64));
    if (lane == 0 && ml != -INFINITY)
      unsafeAtomicMax(p.max_logits + h, ml);
  }
  // ---- epilogue: fp8 sum is scaled by 2^MAX_OFFSET relative to exp2(t-m) ----
  const float lse_new = (l_run > 0.f)
                            ? (m_run - MAX_OFFSET + __log2f(l_run)) *
                                  0.6931471805599453f
                            : -INFINITY;
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  const size_t out_row_pitch = (size_t)p.hq * D;

  const int row_last = min(m0 + FP8_BM, qe) - 1;
  const int s0 = m0 / LOCK_GRAN, s1 = row_last / LOCK_GRAN;
  int* lock0 = p.locks + (size_t)s0 * p.hq + h;
  int* lock1 = p.locks + (size_t)s1 * p.hq + h;
  if (threadIdx.x == 0) {
    int expected = 0;
    while (!__hip_atomic_compare_exchange_strong(
        lock0, &expected, 1, __ATOMIC_ACQUIRE, __ATOMIC_RELAXED,
        __HIP_MEMORY_SCOPE_AGENT))
      expected = 0;
    if (s1 != s0) {
      expected = 0;
      while (!__hip_atomic_compare_exchange_strong(
          lock1, &expected, 1, __ATOMIC_ACQUIRE, __ATOMIC_RELAXED,
          __HIP_MEMORY_SCOPE_AGENT))
        expected = 0;
    }
  }
  __syncthreads();

  float lse_prev = -INFINITY;
  if (qvalid) lse_prev = p.lse[(size_t)qrow * p.hq + h];
  float lse_m;
  {
    const float a = fmaxf(lse_prev, lse_new);
    const float b = fminf(lse_prev, lse_new);
    lse_m = (a == -INFINITY) ? -INFINITY : a + log1pf(expf(b - a));
  }
  const float w_prev = (lse_prev == -INFINITY) ? 0.f : expf(lse_prev - lse_m);
  const float w_new = (lse_new == -INFINITY) ? 0.f : expf(lse_new - lse_m) * inv_l;

  if (qvalid && lse_m != -INFINITY && hi == 0)
    p.lse[(size_t)qrow * p.hq + h] = lse_m;

#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int src = crow(r, hi);
    const float wp = __uint_as_float(
        __builtin_amdgcn_ds_bpermute(src << 2, __float_as_uint(w_prev)));
    const float wn = __uint_as_float(
        __builtin_amdgcn_ds_bpermute(src << 2, __float_as_uint(w_new)));
    const int qr = q0 + src;
    if (qr >= qe || (wp == 0.f && wn == 0.f)) continue;
    const size_t base = (size_t)qr * out_row_pitch + (size_t)h * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      float* ptr = p.out_f32 + base + dt * 32 + lo32;
      const float prev = (wp > 0.f) ? wp * (*ptr) : 0.f;
      *ptr = prev + wn * acc_o[dt][r];
    }
  }

  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    if (s1 != s0)
      __hip_atomic_store(lock1, 0, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
    __hip_atomic_store(lock0, 0, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
  }
}

extern "C" int magi_ffa_fwd_fp8(const magi_ffa_fwd_args* a) {
  if (!a || !a->q || !a->k || !a->v || !a->out || !a->lse) return -1;
  if (a->d != 128 && a->d != 64) return -2;
  if (a->hq % a->hk != 0) return -3;
  if (!a->out_is_fp32 || a->disable_atomic_reduction) return -6;
  if (a->softcap != 0.f) return -7;  // fp8 softcap: later round
  if (a->n_ranges <= 0) return 0;
  if (!a->locks) return -4;
  if (a->n_ranges > 65535) return -5;

  Fp8FwdParams p{};
  p.q = (const u8*)a->q;
  p.k = (const u8*)a->k;
  p.v = (const u8*)a->v;
  p.out_f32 = (float*)a->out;
  p.lse = a->lse;
  p.q_ranges = a->q_ranges;
  p.k_ranges = a->k_ranges;
  p.attn_type_map = a->attn_type_map;
  p.locks = a->locks;
  p.max_logits = a->max_logits;
  // fp8 fwd: auto_range_merge segments not supported yet
  if (a->qk_starts) return -7;
  p.hq = a->hq;
  p.hk = a->hk;
  p.gqa = a->hq / a->hk;
  p.scale = a->softmax_scale;
  p.total_q = a->total_q;
  p.total_k = a->total_k;

  const int mblocks = (a->max_seqlen_q + FP8_BM - 1) / FP8_BM;
  dim3 grid(mblocks, (unsigned)a->n_ranges, a->hq), block(256);
  hipStream_t s = (hipStream_t)a->stream;
  if (a->d == 128)
    hipLaunchKernelGGL((ffa_fwd_fp8_kernel<128>), grid, block, 0, s, p);
  else
    hipLaunchKernelGGL((ffa_fwd_fp8_kernel<64>), grid, block, 0, s, p);
  return (int)hipGetLastError();
}
