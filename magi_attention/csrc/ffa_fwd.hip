// ffa_fwd.hip — MI355X-native flex-flash-attention FORWARD kernel (gfx950).
//
// Re-designed from scratch for CDNA4 (NOT a port of the reference SM90 kernel):
//   - wave64, MFMA v_mfma_f32_32x32x16_bf16 tiles
//   - swapped QK^T (mfma(K, Q)) so each lane owns the scores of ONE q column
//     -> softmax row-reduce is 15 in-register fmax + one shfl_xor(32)
//   - P -> bf16 A-fragments rebuilt in-register via cvt_pk + permlane32_swap
//     (no LDS round trip for P)
//   - fp32 out accumulator merged in gmem under 2-slot range locks
//     (replaces the reference's atomicCAS range-lock epilogue,
//      epilogue_fwd.hpp:271-424), so overlapping q_ranges and cross-launch
//     accumulation (the CP runtime's out_acc) both work.
//
// Numerics mirror the reference (softmax.h:150-331): base-2 exponentials with
// softmax_scale_log2 = scale*log2(e); lse = (m + log2(l)) * ln2 natural-log;
// softcap: score = tanh(s*scale/softcap), softmax scale becomes softcap
// (mainloop_fwd_sm90_tma_gmma_ws.hpp:466-489); empty rows lse=-inf, out=0.
//
// Mask semantics (flex_flash_attn.py:1247-1341), in GLOBAL coords for slice
// (q_range=[qs,qe), k_range=[ks,ke)):
//   causal      : k - q <= ke - qe   (bottom-right aligned)
//   inv_causal  : k - q >= ks - qs   (top-left aligned)
//   bi_causal   : both

#include <hip/hip_runtime.h>
#include <math.h>

#include "../../include/magi_ffa.h"

#define FFA_BM 128   // q rows per workgroup (4 waves x 32)
#define FFA_BN 32    // k rows per inner tile
#define LOCK_GRAN 128

using f32x16 = __attribute__((ext_vector_type(16))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using bf16_t = __bf16;

#define DEV_INLINE __device__ __forceinline__

// Software-pipelined LDS-DMA barrier (see ffa_bwd.hip): wait until at most
// VM vector-memory ops are outstanding (the one prefetch stage this wave
// still has in flight), instead of the full vmcnt(0) drain __syncthreads()
// would insert. Valid because the k loop issues NO other vector-memory ops.
template <int VM>
DEV_INLINE void fwd_pipe_barrier() {
  asm volatile("s_waitcnt vmcnt(%0)\n\ts_waitcnt lgkmcnt(0)\n\ts_barrier"
               ::"i"(VM) : "memory");
}

DEV_INLINE float warp_xor32(float v) { return __shfl_xor(v, 32, 64); }

// one packed convert (RNE, same as the scalar bf16 cast) — the C version
// lowers to 2 cvt + shift + or, 4x the issue slots (guide: cvt_pk idiom)
DEV_INLINE unsigned pack_bf16_pair(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// raw v_exp_f32: the libm exp2f expands to ~5 instructions of denormal-range
// guards; specials (inf/NaN) behave identically, only sub-denormal precision
// differs (P < 1e-38 ~ 0)
DEV_INLINE float fast_exp2(float x) { return __builtin_amdgcn_exp2f(x); }

// C/D fragment row for v_mfma_f32_32x32x16_bf16: reg r, lane-half hi
DEV_INLINE int crow(int r, int hi) { return (r & 3) + 8 * (r >> 2) + 4 * hi; }

// ds_read_b64_tr_b16 pair -> one MFMA B-fragment (see ffa_bwd.hip for the
// probed semantics).
// r2: the r1 inline-asm version forced `s_waitcnt lgkmcnt(0)` + a sched
// fence per fragment — a full ~50-cycle LDS-latency park before every second
// MFMA of the output matmuls. The clang builtin is the same instruction but
// scheduler-visible: reads pipeline across fragments and waits batch.
DEV_INLINE bf16x8 tr16_frag(int a0, int a1) {
  typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4_;
  bf16x4_ v0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4_*)(unsigned)a0);
  bf16x4_ v1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4_*)(unsigned)a1);
  union {
    bf16x4_ h[2];
    bf16x8 v;
  } r;
  r.h[0] = v0;
  r.h[1] = v1;
  return r.v;
}

struct FwdParams {
  const bf16_t* q;
  const bf16_t* k;
  const bf16_t* v;
  float* out_f32;     // used when OUT_BF16==false
  bf16_t* out_bf16;   // used when OUT_BF16==true
  float* lse;
  const int* q_ranges;
  const int* k_ranges;
  const int* attn_type_map;
  int* locks;
  float* max_logits;
  const int* qk_starts;  // auto_range_merge: ri = unique q range; k segments
                         // are k_ranges[qk_starts[ri]..qk_starts[ri+1]) with
                         // per-segment attn types; NULL = one segment per ri
  int hq, hk, gqa;    // gqa = hq / hk
  int head_major;     // 1: grid.x = head (XCD-affine; per-head KV fits L2)
  int work_nx, work_ny, work_nz;  // flattened work space (strided-grid mode)
  int n_lock_slots;
  float scale;        // softmax_scale
  float softcap;
  long long total_q, total_k;
};

// WAVES: q-tiles per workgroup sharing ONE staged K/V image (see the bwd
// kernels): 8 waves (one 512-thread WG/CU = 2 waves/SIMD) halve the staging
// and barrier cost per MFMA for long ranges; 4 for short ranges.
template <int D, bool HAS_SOFTCAP, bool ATOMIC, bool OUT_BF16, int WAVES,
          int NBUF = 2>
__global__ __launch_bounds__(64 * WAVES, 8 / WAVES) void ffa_fwd_kernel(FwdParams p) {
  constexpr int DF = D / 16;    // # of 16-wide d fragments
  constexpr int DT = D / 32;    // # of 32-wide output d tiles
  // Adaptive grid: head-major (blockIdx.x = head -> one XCD per head: L2
  // locality for K/V and merge traffic) when one head's K/V fits an XCD's
  // 4 MB L2; m-block-major otherwise (all XCDs stream the same K window).
  // CU-margin (reference env/comm.py:44 sm_margin): when cu_margin > 0 the
  // launcher caps the grid below the chip's resident-WG count and each WG
  // walks the flattened work space, leaving `margin` CUs permanently free
  // for comm kernels. With margin 0 the grid covers the work exactly and
  // this loop runs once per WG (identical to the r1 3D grid).
  const long long work_total =
      (long long)p.work_nx * p.work_ny * p.work_nz;
  const long long grid_span = (long long)gridDim.x * gridDim.y * gridDim.z;
  for (long long w = blockIdx.x +
           (long long)gridDim.x * (blockIdx.y + (long long)gridDim.y * blockIdx.z);
       w < work_total; w += grid_span) {
  const int wx = (int)(w % p.work_nx);
  const long long w2 = w / p.work_nx;
  const int wy = (int)(w2 % p.work_ny);
  const int wz = (int)(w2 / p.work_ny);
  const int ri = p.head_major ? wz : wy;
  const int h = p.head_major ? wx : wz;
  const int mb = p.head_major ? wy : wx;
  const int qs = p.q_ranges[2 * ri], qe = p.q_ranges[2 * ri + 1];
  const int m0 = qs + mb * (32 * WAVES);
  if (m0 >= qe) continue;                     // uniform across block
  // auto_range_merge: iterate this unique q range's k segments in-kernel,
  // online softmax carried across them (reference merge_range.cu semantics)
  const int seg0 = p.qk_starts ? p.qk_starts[ri] : ri;
  const int seg1 = p.qk_starts ? p.qk_starts[ri + 1] : ri + 1;
  int ks, ke, atype;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  // static priority for the second-dispatched half of an 8-wave WG: the
  // younger waves are the VALU-arbitration losers on every segment
  // (MI355X guide, two-waves-per-SIMD item 4)
  if (WAVES == 8 && wave >= 4) asm volatile("s_setprio 1");

  const int lo32 = lane & 31;
  const int hi = lane >> 5;

  const int q0 = m0 + wave * 32;
  const int qrow = q0 + lo32;
  const bool qvalid = (qrow < qe) && (q0 < qe);
  const bool qvalid_any = q0 < qe;
  const int qclamp = qvalid ? qrow : (qe - 1);

  // softmax scales (reference mainloop_fwd...hpp:466-489)
  const float sl2 = HAS_SOFTCAP ? p.softcap * 1.4426950408889634f
                                : p.scale * 1.4426950408889634f;
  const float cap_pre = HAS_SOFTCAP ? p.scale / p.softcap : 0.f;

  // per-wave / block k loop bounds (recomputed per k segment)
  int n_lo = 0, n_hi = 0, b_lo = 0, b_hi = 0;
  auto seg_bounds = [&]() {
    n_lo = ks; n_hi = ke;
    if (ke > ks && q0 < qe) {
      const int qhiw = min(q0 + 31, qe - 1);
      if (atype == 1 || atype == 3) n_hi = min(n_hi, qhiw + (ke - qe) + 1);
      if (atype == 2 || atype == 3) n_lo = max(n_lo, q0 + (ks - qs));
    } else {
      n_hi = n_lo;  // empty
    }
    b_lo = ks; b_hi = ke;
    if (ke > ks && m0 < qe) {
      const int qhib = min(m0 + 32 * WAVES - 1, qe - 1);
      if (atype == 1 || atype == 3) b_hi = min(b_hi, qhib + (ke - qe) + 1);
      if (atype == 2 || atype == 3) b_lo = max(b_lo, m0 + (ks - qs));
    } else {
      b_hi = b_lo;
    }
  };
  // LDS rows are padded to a power of two so the XOR swizzle algebra holds
  // for every D (D=192 pads 384-byte rows to 512; the slot permutation is a
  // bijection, so padding slots only ever hold data no read requests).
  constexpr int ROWB = (D == 192) ? 512 : D * 2;
  constexpr int ROWE = ROWB / 2;       // LDS elements per padded row
  constexpr int VSLOTS = D / 8;        // valid 16-B source slots per row
  // 32-B-granular swizzle: b128 lane groups spread (<=2-way) AND every 32-B
  // run stays physically contiguous for the tr16 V reads (see bwd kernels)
  constexpr int SW32M = ROWB / 32 - 1;
  auto swz = [](int row, int byte_off) {
    return byte_off ^ ((row & SW32M) << 5);
  };
  // NBUF x 64-row K/V images staged by LDS-DMA (same scheme as the
  // backward kernels): one barrier per 64 k rows, the doubled compute phase
  // covers the prefetch latency the 32-row register pipeline could not.
  // NBUF=3 (r2, ported from the bwd 3-ring): constant-distance vmcnt wait —
  // the barrier waits only for the one stage it needs and the prefetch gets
  // two compute phases to land, removing the full vmcnt(0) drain per
  // iteration (the dominant stall on short ranges / 8k).
  constexpr int KITER = 2 * FFA_BN;
  __shared__ __attribute__((aligned(16))) char fsmem[NBUF * 2 * KITER * ROWB];
  auto lds_k = [&](int buf) -> __bf16* {
    return (__bf16*)(fsmem + (2 * buf) * KITER * ROWB);
  };
  auto lds_v = [&](int buf) -> __bf16* {
    return (__bf16*)(fsmem + (2 * buf + 1) * KITER * ROWB);
  };

  // Q fragments in registers (8 x bf16x8 for D=128)
  bf16x8 qf[DF];
  {
    const bf16_t* qp = p.q + (size_t)qclamp * p.hq * D + (size_t)h * D;
#pragma unroll
    for (int dd = 0; dd < DF; ++dd)
      qf[dd] = *(const bf16x8*)(qp + dd * 16 + hi * 8);
  }

  const int kh = h / p.gqa;
  const size_t k_pitch = (size_t)p.hk * D;
  const bf16_t* kbase = p.k + (size_t)kh * D;
  const bf16_t* vbase = p.v + (size_t)kh * D;

  float m_run = -INFINITY;
  float l_run = 0.f;
  f32x16 acc_o[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) acc_o[dt] = (f32x16)(0.f);

  constexpr int ROWS_PER_GLDS = 1024 / ROWB;
  static_assert(KITER / WAVES >= ROWS_PER_GLDS, "stage rows per wave");
  constexpr int GLDS_PER_WAVE = (KITER / WAVES) / ROWS_PER_GLDS;
  constexpr int GOPS = 2 * GLDS_PER_WAVE;  // vm ops per stage call per wave
  auto stage_glds = [&](int buf, int n0x) {
#pragma unroll
    for (int gi = 0; gi < GLDS_PER_WAVE; ++gi) {
      const int r0 = (KITER / WAVES) * wave + ROWS_PER_GLDS * gi;
      const int r = r0 + lane / (ROWB / 16);
      const int c = lane % (ROWB / 16);
      const int kr = min(n0x + r, ke - 1);
      int cs = c ^ ((r & SW32M) << 1);
      if (cs >= VSLOTS) cs = 0;  // padding slot: any in-bounds source
      const int csw = cs * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              kbase + (size_t)kr * k_pitch + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_k(buf)[r0 * ROWE],
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              vbase + (size_t)kr * k_pitch + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_v(buf)[r0 * ROWE],
          16, 0, 0);
    }
  };

  int cur = 0;

  auto sub_body = [&](int ns, const __bf16* lkb, const __bf16* lvb) {
    const bool live = (ns + FFA_BN > n_lo) && (ns < n_hi) && qvalid_any;
    if (!live) return;

    f32x16 s = (f32x16)(0.f);
    {
#pragma unroll
      for (int dd = 0; dd < DF; ++dd) {
        bf16x8 kf = *(const bf16x8*)(
            (const char*)lkb + swz(lo32, lo32 * ROWB + dd * 32 + hi * 16));
        s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[dd], s, 0, 0, 0);
      }
    }

    // ---- mask + scale into exp2 domain ----
    // interior fast path: when the whole tile is provably unmasked for every
    // lane (wave-uniform), skip the per-element compare/select chain — the
    // kernels are instruction-issue-bound and masks only bind near edges.
    const bool interior =
        (q0 + 31 < qe) && (ns >= n_lo) && (ns + FFA_BN <= n_hi) &&
        !((atype == 1 || atype == 3) && (ns + FFA_BN - 1 > q0 + (ke - qe))) &&
        !((atype == 2 || atype == 3) && (ns < q0 + 31 + (ks - qs)));
    float t[16];
    float mx = -INFINITY;
    if (interior) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float sv = s[r];
        if (HAS_SOFTCAP) sv = tanhf(sv * cap_pre);
        t[r] = sv * sl2;
        mx = fmaxf(mx, t[r]);
      }
    } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kk = ns + crow(r, hi);
        bool ok = qvalid && (kk >= n_lo) && (kk < n_hi);
        if (atype == 1 || atype == 3) ok = ok && (kk - qrow <= ke - qe);
        if (atype == 2 || atype == 3) ok = ok && (kk - qrow >= ks - qs);
        float sv = s[r];
        if (HAS_SOFTCAP) sv = tanhf(sv * cap_pre);
        t[r] = ok ? sv * sl2 : -INFINITY;
        mx = fmaxf(mx, t[r]);
      }
    }
    mx = fmaxf(mx, warp_xor32(mx));

    // ---- defer-max (guide RESCALE_THRESHOLD idiom): tolerate per-row max
    // growth up to 2^8 without touching m_run — P stays bounded by 256 in
    // the fp32 accumulators and the O-rescale (16 bpermute + 64 mults)
    // vanishes on the vast majority of tiles. A lane whose first finite max
    // arrives (m_run -inf -> finite) cannot defer. T13 hazard: the decision
    // is made ONCE per tile before PV, and the same alpha feeds both l_run
    // and the O rescale.
    const float m_new = fmaxf(m_run, mx);
    // max_logits reads m_run as the TRUE row max in the epilogue — no slack
    const float defer_thr = p.max_logits ? 0.f : 8.f;
    const bool need_rescale =
        (m_new > m_run + defer_thr) ||
        (m_run == -INFINITY && m_new != -INFINITY);
    float m_use;
    float alpha;
    if (!__any(need_rescale)) {
      m_use = (m_run == -INFINITY) ? 0.f : m_run;  // keep the old reference
      alpha = 1.f;
    } else {
      m_use = (m_new == -INFINITY) ? 0.f : m_new;
      alpha = (m_run == -INFINITY) ? 0.f : fast_exp2(m_run - m_use);
      m_run = m_new;
    }

    float pr[16];
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      pr[r] = fast_exp2(t[r] - m_use);   // exp2(-inf)=0 for masked
      psum += pr[r];
    }
    l_run = l_run * alpha + (psum + warp_xor32(psum));

    // ---- rescale O by alpha (per q = crow layout) ----
    // NOTE: wave-uniform condition — ds_bpermute reads other lanes' registers,
    // so every lane must be active whenever any lane needs the rescale.
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

    // ---- P -> bf16 A-fragments (cvt_pk + permlane32_swap) ----
    bf16x8 pa[2];
#pragma unroll
    for (int tt = 0; tt < 2; ++tt) {
      unsigned c0 = pack_bf16_pair(pr[8 * tt + 0], pr[8 * tt + 1]);
      unsigned c1 = pack_bf16_pair(pr[8 * tt + 2], pr[8 * tt + 3]);
      unsigned c2 = pack_bf16_pair(pr[8 * tt + 4], pr[8 * tt + 5]);
      unsigned c3 = pack_bf16_pair(pr[8 * tt + 6], pr[8 * tt + 7]);
      {
        auto r2 = __builtin_amdgcn_permlane32_swap(c0, c2, false, false);
        c0 = r2[0];
        c2 = r2[1];
      }
      {
        auto r2 = __builtin_amdgcn_permlane32_swap(c1, c3, false, false);
        c1 = r2[0];
        c3 = r2[1];
      }
      union {
        unsigned u[4];
        bf16x8 v;
      } cvt;
      cvt.u[0] = c0;
      cvt.u[1] = c1;
      cvt.u[2] = c2;
      cvt.u[3] = c3;
      pa[tt] = cvt.v;
    }

    // ---- PV: B-frags via ds_read_b64_tr_b16 off the V row image ----
    {
      const int qhalf = (lane >> 4) & 1;
      const int jrow = (lane & 15) >> 2;
      const int v_base = (int)(unsigned long long)(
          (__attribute__((address_space(3))) const char*)lvb);
      const int lane8 = (lane & 3) * 8;
#pragma unroll
      for (int tt = 0; tt < 2; ++tt) {
        const int row0 = 16 * tt + 8 * hi + jrow;
        const int row1 = row0 + 4;
        const int sw0 = (row0 & SW32M) << 5;
        const int sw1 = (row1 & SW32M) << 5;
        const int rb0 = v_base + row0 * ROWB + lane8;
        const int rb1 = v_base + row1 * ROWB + lane8;
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) {
          const int dcol = (dt * 32 + 16 * qhalf) * 2;
          bf16x8 bv = tr16_frag(rb0 + (dcol ^ sw0), rb1 + (dcol ^ sw1));
          acc_o[dt] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[tt], bv, acc_o[dt], 0, 0, 0);
        }
      }
    }
  };

  bool any_seg = false;
  for (int seg = seg0; seg < seg1; ++seg) {
    ks = p.k_ranges[2 * seg];
    ke = p.k_ranges[2 * seg + 1];
    atype = p.attn_type_map ? p.attn_type_map[seg] : 0;
    if (ke <= ks) continue;  // uniform across block
    seg_bounds();
    if (b_lo >= b_hi) continue;
    any_seg = true;
    cur = 0;
    stage_glds(0, b_lo);
    if constexpr (NBUF == 3) stage_glds(1, b_lo + KITER);  // rows clamp
    for (int n0 = b_lo; n0 < b_hi; n0 += KITER) {
      // NBUF=3: wait ONLY for buf[cur] (leave the next stage in flight),
      // then prefetch two stages ahead — always, with clamped rows, so the
      // vmcnt distance stays constant. NBUF=2: full drain.
      fwd_pipe_barrier<NBUF == 3 ? GOPS : 0>();
      if constexpr (NBUF == 3) {
        stage_glds(cur == 0 ? 2 : cur - 1, n0 + 2 * KITER);
      } else {
        if (n0 + KITER < b_hi) stage_glds(cur ^ 1, n0 + KITER);
      }
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int ns = n0 + sub * FFA_BN;
        if (ns >= b_hi) break;  // b_hi is block-uniform
        sub_body(ns, lds_k(cur) + sub * FFA_BN * ROWE,
                 lds_v(cur) + sub * FFA_BN * ROWE);
      }
      cur = (NBUF == 3) ? (cur == 2 ? 0 : cur + 1) : (cur ^ 1);
    }
    // LDS reads of this segment retired before the next segment's staging
    // (implicit vmcnt(0) drains the in-flight prefetches too)
    __syncthreads();
  }
  if (any_seg) {

  // ======================= epilogue =======================
  if (p.max_logits) {
    // per-head max of the scaled (softcapped) logits; m_run is the row max
    // in the exp2 domain -> x ln2. Empty rows are -inf and drop out.
    float ml = m_run * 0.6931471805599453f;
#pragma unroll
    for (int off = 32; off; off >>= 1)
      ml = fmaxf(ml, __shfl_xor(ml, off, 64));
    if (lane == 0 && ml != -INFINITY)
      unsafeAtomicMax(p.max_logits + h, ml);
  }
  const float lse_new =
      (l_run > 0.f) ? (m_run + __log2f(l_run)) * 0.6931471805599453f : -INFINITY;
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;

  const size_t out_row_pitch = (size_t)p.hq * D;

  if (!ATOMIC) {
    // direct store: q_ranges guaranteed disjoint; rows with l==0 keep initial
    if (l_run > 0.f && qvalid && hi == 0)
      p.lse[(size_t)qrow * p.hq + h] = lse_new;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int src = crow(r, hi);
      const float ilq = __uint_as_float(
          __builtin_amdgcn_ds_bpermute(src << 2, __float_as_uint(inv_l)));
      const int qr = q0 + src;
      const bool valid = (qr < qe) && ilq > 0.f;
      if (!valid) continue;
      const size_t base = (size_t)qr * out_row_pitch + (size_t)h * D;
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        const float val = acc_o[dt][r] * ilq;
        if (OUT_BF16)
          p.out_bf16[base + dt * 32 + lo32] = (bf16_t)val;
        else
          p.out_f32[base + dt * 32 + lo32] = val;
      }
    }
  } else {

  // ---- lock-guarded read-merge-write (range locks; a 32*WAVES-row block
  // spans up to 32*WAVES/LOCK_GRAN+1 slots, acquired in ascending order —
  // the global order prevents deadlock between overlapping blocks) ----
  const int row_last = min(m0 + 32 * WAVES, qe) - 1;
  const int s0 = m0 / LOCK_GRAN, s1 = row_last / LOCK_GRAN;
  if (threadIdx.x == 0) {
    for (int sl = s0; sl <= s1; ++sl) {
      int* lk = p.locks + (size_t)sl * p.hq + h;
      int expected = 0;
      while (!__hip_atomic_compare_exchange_strong(
          lk, &expected, 1, __ATOMIC_ACQUIRE, __ATOMIC_RELAXED,
          __HIP_MEMORY_SCOPE_AGENT))
        expected = 0;
    }
  }
  __syncthreads();

  // merged lse per lane (q = lo32 layout)
  float lse_prev = -INFINITY;
  if (qvalid) lse_prev = p.lse[(size_t)qrow * p.hq + h];
  float lse_m;
  {
    const float a = fmaxf(lse_prev, lse_new);
    const float b = fminf(lse_prev, lse_new);
    lse_m = (a == -INFINITY) ? -INFINITY : a + log1pf(expf(b - a));
  }
  const float w_prev = (lse_prev == -INFINITY) ? 0.f : expf(lse_prev - lse_m);
  // fold 1/l into the new weight
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
      // first writer (wp==0) must not read: `out` may be uninitialised
      const float prev = (wp > 0.f) ? wp * (*ptr) : 0.f;
      *ptr = prev + wn * acc_o[dt][r];
    }
  }

  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    for (int sl = s1; sl >= s0; --sl)
      __hip_atomic_store(p.locks + (size_t)sl * p.hq + h, 0, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_AGENT);
  }
  }  // ATOMIC epilogue
  }  // any_seg
  }  // strided work loop
}

// ------------------------------------------------------------------
// MFMA layout probe (GPU self-test; see tests/test_ffa_gpu.py)
// ------------------------------------------------------------------
__global__ void probe_mfma_kernel(const bf16_t* a, const bf16_t* b, float* d) {
  const int lane = threadIdx.x & 63;
  const int lo32 = lane & 31, hi = lane >> 5;
  bf16x8 af = *(const bf16x8*)(a + lo32 * 16 + hi * 8);  // A[32][16] row-major
  bf16x8 bf;
  union {
    unsigned short u[8];
    bf16x8 v;
  } bu;
  for (int e = 0; e < 8; ++e)
    bu.u[e] = *(const unsigned short*)(b + (hi * 8 + e) * 32 + lo32);  // B[16][32]
  bf = bu.v;
  f32x16 acc = (f32x16)(0.f);
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
  for (int r = 0; r < 16; ++r) d[(size_t)crow(r, hi) * 32 + lo32] = acc[r];
}

// probe permlane32_swap + ds_bpermute semantics: in[64] u32 per lane.
// out0/out1 = the two results of permlane32_swap(in, in2) where in2 = in+1000;
// out2 = ds_bpermute(addr = (lane%32)<<2, in).
__global__ void probe_lane_kernel(const unsigned* in, unsigned* out) {
  const int lane = threadIdx.x & 63;
  unsigned a = in[lane];
  unsigned b = in[lane] + 1000u;
  auto r2 = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  out[lane] = r2[0];
  out[64 + lane] = r2[1];
  out[128 + lane] =
      __builtin_amdgcn_ds_bpermute((lane & 31) << 2, in[lane]);
}

extern "C" int magi_probe_lane(const void* in, void* out, void* stream) {
  hipLaunchKernelGGL(probe_lane_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const unsigned*)in, (unsigned*)out);
  return (int)hipGetLastError();
}

extern "C" int magi_probe_mfma(const void* a, const void* b, void* d,
                               void* stream) {
  hipLaunchKernelGGL(probe_mfma_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const bf16_t*)a, (const bf16_t*)b,
                     (float*)d);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------------
// launcher
// ------------------------------------------------------------------
template <int D, int W, int NB>
static int launch_fwd_d(const magi_ffa_fwd_args* a, const FwdParams& p,
                        dim3 grid, dim3 block, hipStream_t stream) {
  const bool sc = a->softcap > 0.f;
  const bool atomic = !a->disable_atomic_reduction;
  const bool obf16 = !a->out_is_fp32;
  if (atomic && obf16) return -10;  // atomic merge requires fp32 out
#define LAUNCH(SC, AT, OB)                                                 \
  hipLaunchKernelGGL((ffa_fwd_kernel<D, SC, AT, OB, W, NB>), grid, block,  \
                     0, stream, p)
  if (atomic) {
    if (sc) LAUNCH(true, true, false);
    else LAUNCH(false, true, false);
  } else if (obf16) {
    if (sc) LAUNCH(true, false, true);
    else LAUNCH(false, false, true);
  } else {
    if (sc) LAUNCH(true, false, false);
    else LAUNCH(false, false, false);
  }
#undef LAUNCH
  return (int)hipGetLastError();
}

extern "C" int magi_ffa_fwd(const magi_ffa_fwd_args* a) {
  if (!a || !a->q || !a->k || !a->v || !a->out || !a->lse) return -1;
  if (a->d != 64 && a->d != 128 && a->d != 192) return -2;
  if (a->hq % a->hk != 0) return -3;
  if (a->n_ranges <= 0) return 0;
  if (!a->disable_atomic_reduction && !a->locks) return -4;

  FwdParams p;
  p.q = (const bf16_t*)a->q;
  p.k = (const bf16_t*)a->k;
  p.v = (const bf16_t*)a->v;
  p.out_f32 = (float*)a->out;
  p.out_bf16 = (bf16_t*)a->out;
  p.lse = a->lse;
  p.q_ranges = a->q_ranges;
  p.k_ranges = a->k_ranges;
  p.attn_type_map = a->attn_type_map;
  p.locks = a->locks;
  p.max_logits = a->max_logits;
  p.qk_starts = a->qk_starts;
  p.hq = a->hq;
  p.hk = a->hk;
  p.gqa = a->hq / a->hk;
  p.n_lock_slots = (int)((a->total_q + LOCK_GRAN - 1) / LOCK_GRAN);
  p.scale = a->softmax_scale;
  p.softcap = a->softcap;
  p.total_q = a->total_q;
  p.total_k = a->total_k;

  // r2: 8 waves (one 512-thread WG/CU) now beat 4 on both 8k (505 vs 471
  // TF) and 64k (742 vs 716) — the r1 preference for 4 predated the
  // scheduler-visible tr16 reads, packed converts and raw-exp2 softmax.
  int fw = 8;
  { const char* e = getenv("MAGI_FWD_WAVES"); if (e && atoi(e)) fw = atoi(e); }
  if (a->d == 192) fw = 4;  // the W8 D=192 build spills (256-reg cap)
  const int span = 32 * fw;
  const int mblocks = (a->max_seqlen_q + span - 1) / span;
  if (a->n_ranges > 65535) return -5;
  // head-major XCD affinity (grid.x = head): late-r2 A/B moved the
  // threshold from 4 MB to 16 MB of per-head K+V — head-major wins 1.26x
  // at 4k, 1.45x at 8k, 1.29x at 16k (8.4 MB), is neutral at 32k
  // (16.8 MB) and loses 0.93x at 64k where the streamed K window
  // L3-shares better work-major (profiles/r2_headmajor_ab.md)
  p.head_major = ((long long)a->total_k * a->d * 4 <= (16 << 20)) ? 1 : 0;
  { const char* e = getenv("MAGI_FWD_HEADMAJOR");  // A/B override
    if (e) p.head_major = atoi(e) ? 1 : 0; }
  dim3 grid = p.head_major
                  ? dim3(a->hq, mblocks, (unsigned)a->n_ranges)
                  : dim3(mblocks, (unsigned)a->n_ranges, a->hq);
  p.work_nx = grid.x;
  p.work_ny = grid.y;
  p.work_nz = grid.z;
  const int margin = a->cu_margin & 0xFFFF;
  if (margin > 0) {
    // persistent-strided: cap resident WGs to (256 - margin) CUs
    const long long total = (long long)grid.x * grid.y * grid.z;
    const int per_cu = (fw == 4) ? 2 : 1;
    long long cap = (long long)(256 - margin) * per_cu;
    if (cap < 1) cap = 1;
    if (cap > total) cap = total;
    grid = dim3((unsigned)cap, 1, 1);
  }
  dim3 block(64 * fw);
  hipStream_t stream = (hipStream_t)a->stream;
  // 3-slot staging ring (constant-distance vmcnt barrier) for the W8 path;
  // the W4/D=192 builds keep NBUF=2 (a 3-ring of 512-B rows exceeds LDS,
  // and W4 runs 2 WGs/CU which the 96 KB ring would halve)
  int nbuf = 3;
  { const char* e = getenv("MAGI_FWD_NBUF"); if (e) nbuf = atoi(e); }
  if (a->d == 64)
    return fw == 8 ? (nbuf == 3
                          ? launch_fwd_d<64, 8, 3>(a, p, grid, block, stream)
                          : launch_fwd_d<64, 8, 2>(a, p, grid, block, stream))
                   : launch_fwd_d<64, 4, 2>(a, p, grid, block, stream);
  if (a->d == 192)
    return fw == 8 ? launch_fwd_d<192, 8, 2>(a, p, grid, block, stream)
                   : launch_fwd_d<192, 4, 2>(a, p, grid, block, stream);
  return fw == 8 ? (nbuf == 3
                        ? launch_fwd_d<128, 8, 3>(a, p, grid, block, stream)
                        : launch_fwd_d<128, 8, 2>(a, p, grid, block, stream))
               : launch_fwd_d<128, 4, 2>(a, p, grid, block, stream);
}

extern "C" int magi_ffa_abi_version(void) { return 1; }

// ------------------------------------------------------------------
// probe ds_read_b64_tr_b16 semantics (see tests/gpu_probe_tr16.py)
// ------------------------------------------------------------------
__global__ void probe_tr16_kernel(const unsigned short* in, unsigned short* out,
                                  int mode) {
  __shared__ __attribute__((aligned(16))) unsigned short lds[2048];
  const int lane = threadIdx.x & 63;
  for (int i = lane; i < 2048; i += 64) lds[i] = in[i];
  __syncthreads();
  int addr;  // byte address into LDS
  if (mode == 0) addr = 0;                       // uniform base
  else if (mode == 1) addr = lane * 8;           // lane-linear 8B
  else addr = (lane & 15) * 8 + (lane >> 4) * 128;  // grouped guess
  unsigned long long v;
  const int a = (int)(unsigned long long)(
                    (__attribute__((address_space(3))) char*)lds) +
                addr;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %1\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=v"(v)
      : "v"(a)
      : "memory");
  __builtin_amdgcn_sched_barrier(0);
  for (int j = 0; j < 4; ++j)
    out[lane * 4 + j] = (unsigned short)(v >> (16 * j));
}

extern "C" int magi_probe_tr16(const void* in, void* out, int mode,
                               void* stream) {
  hipLaunchKernelGGL(probe_tr16_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const unsigned short*)in,
                     (unsigned short*)out, mode);
  return (int)hipGetLastError();
}
