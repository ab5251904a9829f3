// ffa_bwd.hip — MI355X-native flex-flash-attention BACKWARD (gfx950), v1.
//
// Behaviour: reference flash_bwd_kernel_sm90.h:41 /
// mainloop_bwd_sm90_tma_gmma_ws.hpp (5-matmul recompute backward, fp32 atomic
// dQ/dK/dV accumulation; preprocess dPsum = rowsum(dO*O),
// flash_bwd_preprocess_kernel.h:42).
//
// Structure v1: ONE workgroup = 4 waves = 128 k rows (each wave one 32-row
// k tile of the same range+head). The workgroup loops over the slice's q
// tiles; per iteration Q[32][D] and dO[32][D] are staged ONCE into
// XOR-swizzled LDS by all 256 threads (coalesced 16-B loads) and consumed by
// all 4 waves — replacing v0's per-wave scalar global loads. Per wave:
//   S^T  = K Q^T ; dP^T = V dO^T      (swapped MFMA, q-side operands from LDS)
//   P^T, dS^T                          (base-2 recompute, like the fwd)
//   dV  += P^T dO ; dK += dS^T Q       (P/dS transposed through per-wave LDS)
//   dQ  += atomicAdd(dS K)             (dS^T -> A-frag in-register, permlane)
// dK/dV accumulate in AGPRs across the q loop, atomic-added once at the end.

#include <hip/hip_runtime.h>
#include <math.h>
#include <stdlib.h>
#include <type_traits>

#include "../../include/magi_ffa.h"

#define BWD_BN 32    // k rows per wave
#define BWD_BM 32    // q rows per tile
#define BWD_WAVES 4  // k tiles per workgroup

using f32x16 = __attribute__((ext_vector_type(16))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using bf16_t = __bf16;

#define DEV_INLINE __device__ __forceinline__

// Software-pipelined LDS-DMA barrier: waits until at most VM vector-memory
// ops are outstanding (= exactly the NEXT prefetch group, leaving the current
// tile's group retired), makes LDS reads visible, then syncs the block.
// vmcnt retires in issue order, so a constant distance works as long as every
// iteration issues the same number of vector-memory ops — stage_glds is
// therefore called unconditionally with clamped addresses past the end.
// __syncthreads() would insert s_waitcnt vmcnt(0) and collapse the pipeline
// to depth one (measured: 34-42%% SQ_WAIT in both backward kernels).
template <int VM>
DEV_INLINE void pipe_barrier() {
  asm volatile("s_waitcnt vmcnt(%0)\n\ts_waitcnt lgkmcnt(0)\n\ts_barrier"
               :: "n"(VM) : "memory");
}

DEV_INLINE int crow(int r, int hi) { return (r & 3) + 8 * (r >> 2) + 4 * hi; }

DEV_INLINE bool wave_alive(int m0, int qe) { return m0 < qe; }

// ds_read_b64_tr_b16 pair -> one MFMA B-fragment (8 bf16).
// Probed semantics (tests/gpu_probe_tr16.py): within each 16-lane group, the
// addresses of lanes =0 (mod 4) give four ROW base addresses; lane l receives
// element (row_j_base + (l&15)) for j=0..3. Two reads cover 8 rows.
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

DEV_INLINE bf16x8 cframe_to_afrag(const float* val, int tt) {
  unsigned c0 = pack_bf16_pair(val[8 * tt + 0], val[8 * tt + 1]);
  unsigned c1 = pack_bf16_pair(val[8 * tt + 2], val[8 * tt + 3]);
  unsigned c2 = pack_bf16_pair(val[8 * tt + 4], val[8 * tt + 5]);
  unsigned c3 = pack_bf16_pair(val[8 * tt + 6], val[8 * tt + 7]);
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
  return cvt.v;
}

struct BwdParams {
  const bf16_t* dout;
  const bf16_t* q;
  const bf16_t* k;
  const bf16_t* v;
  const void* out;
  const float* lse;
  float* dq;
  float* dk;
  float* dv;
  float* dpsum;
  const int* q_ranges;
  const int* k_ranges;
  const int* attn_type_map;
  int hq, hk, gqa;
  float scale;
  float softcap;
  long long total_q, total_k;
  int debug_ablate;  // perf ablation only: 1=skip dq stores, 2=skip dkv stores
  int head_major;    // 1: blockIdx.x = head (XCD-affine); 0: work-major
  int work_nx, work_ny, work_nz;  // flattened work space (strided-grid mode)
  int head_mult;     // deterministic GQA split: real head = off + idx*mult
  int head_off;
  int n_heads_launch;
  const int* seg_starts;  // auto_range_merge: dq pass: ri = unique q range,
                          // k segments; dkv pass: ri = unique k range, q
                          // segments (reference bwd_kq_map). NULL = 1:1.
};

// ---------------- preprocess: dpsum = rowsum(dO * O) ----------------
template <bool OUT_F32>
__global__ __launch_bounds__(256) void bwd_preprocess_kernel(BwdParams p, int d,
                                                             long long n_rows) {
  const long long row = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= n_rows) return;
  const int lane = threadIdx.x & 63;
  const int per = d / 64;
  const bf16_t* do_p = p.dout + row * d;
  float acc = 0.f;
  for (int e = 0; e < per; ++e) {
    const int idx = lane * per + e;
    const float dov = (float)do_p[idx];
    float ov;
    if (OUT_F32)
      ov = ((const float*)p.out)[row * d + idx];
    else
      ov = (float)((const bf16_t*)p.out)[row * d + idx];
    acc += dov * ov;
  }
#pragma unroll
  for (int s = 32; s > 0; s >>= 1) acc += __shfl_xor(acc, s, 64);
  if (lane == 0) p.dpsum[row] = acc;
}


// ---------------- mainloop ----------------
// MODE 0: fused dK+dV — r2 redesign: V fragments live in a per-wave LDS tile
// (staged once per block) instead of 32 persistent VGPRs, which brings the
// fused kernel under the 256-reg budget of 2 waves/SIMD (r1's fused version
// was 437 regs = 1 wave/SIMD, or 65 scratch spills RELOADED EVERY ITERATION
// when forced to 256). Shares S, P, exp2 and the staged Q/dO image between
// the dV and dK matmuls: 32 MFMAs per subtile vs the split modes' 40.
// MODE 1: dV only / MODE 2: dK only — the r1 split, kept selectable.
// WAVES: k-tiles per workgroup sharing ONE staged Q/dO image.
// NBUF: LDS staging ring depth. 2 = r1 behaviour (barrier drains vmcnt(0),
// prefetch has one compute phase to land). 3 = constant-distance vmcnt(G)
// barrier: prefetch has TWO compute phases, and the barrier only waits for
// the one stage it needs — attacks the 34-42% SQ_WAIT_ANY of the r1 PMC.
template <int D, bool HAS_SOFTCAP, int MODE, int WAVES, int NBUF, int NT = 0>
__global__ __launch_bounds__(64 * WAVES, WAVES >= 6 ? 2 : 1)
void ffa_bwd_dkv_kernel(BwdParams p) {
  // NT: stage with the non-temporal policy (aux=2) — staged rows are read
  // once per WG; keeping them out of L2 protects the resident K strips and
  // V tiles from streaming eviction (PMC: 174 GB/launch L2-miss traffic)
  constexpr int STAGE_AUX = NT ? 2 : 0;
  constexpr bool WANT_DV = MODE != 2;
  constexpr bool WANT_DK = MODE != 1;
  // MODE 3 = fused "fat wave" (r2-v3): 4 waves x 64 k rows (2 column tiles)
  // at 1 wave/SIMD — the unified 512-reg file holds BOTH columns' dK/dV
  // accumulators (256 regs -> AGPRs), 64 MFMAs per wave between barriers
  // instead of 32, and half the waves to sync per barrier. Same 256-row
  // k span per WG as MODE 0 at 8 waves, identical grid.
  constexpr bool V_IN_LDS = (MODE == 0 || MODE == 3);
  constexpr int NCOL = (MODE == 3) ? 2 : 1;
  constexpr int DF = D / 16;
  constexpr int DT = D / 32;
  // LDS rows padded to a power of two (D=192 -> 512-byte rows; see
  // ffa_fwd.hip note on the swizzle bijection)
  constexpr int ROWB = (D == 192) ? 512 : D * 2;
  constexpr int ROWE = ROWB / 2;
  constexpr int VSLOTS = D / 8;
  // XOR swizzle within one LDS row: spreads a b128 lane group over the row's
  // 16-B slots (T2/G4); mask keeps the XOR inside the row for D=64 too.
  // 32-B-granular XOR swizzle: spreads b128 lane groups over the row's 16-B
  // slot pairs (<=2-way) while keeping every 32-B run physically contiguous,
  // which the tr16 row reads require; mask scales with row size (D=64 rows
  // are 128 B).
  constexpr int SW32M = ROWB / 32 - 1;
  auto swz = [](int row, int byte_off) {
    return byte_off ^ ((row & SW32M) << 5);
  };
  // Adaptive grid (see fwd kernel): head-major = XCD-affine atomics + K/V
  // L2 locality per head; work-major when one head's operands overflow an
  // XCD's L2.
  // CU-margin strided-grid mode (see ffa_fwd.hip): margin>0 caps the grid
  // below the resident-WG count; each WG walks the flattened work space.
  // MODE 0 sits at exactly the 256-reg budget and the loop backedge costs it
  // 41 scratch spills — its launcher never caps the grid, so the loop is
  // made provably single-iteration there and dissolves at compile time.
  const long long work_total =
      (long long)p.work_nx * p.work_ny * p.work_nz;
  const long long grid_span = (long long)gridDim.x * gridDim.y * gridDim.z;
  const long long w_first = blockIdx.x +
      (long long)gridDim.x * (blockIdx.y + (long long)gridDim.y * blockIdx.z);
  for (long long w = w_first; w < work_total; w += grid_span) {
  const int wx = (int)(w % p.work_nx);
  const long long w2 = w / p.work_nx;
  const int wy = (int)(w2 % p.work_ny);
  const int wz = (int)(w2 / p.work_ny);
  const int ri = wz;
  const int hidx = p.head_major ? wx : wy;
  const int h = p.head_off + hidx * p.head_mult;
  const int wb = p.head_major ? wy : wx;
  const int ks = p.k_ranges[2 * ri], ke = p.k_ranges[2 * ri + 1];
  const int nblk0 = ks + wb * (BWD_BN * NCOL * WAVES);
  if (nblk0 >= ke) continue;
  const int seg0 = p.seg_starts ? p.seg_starts[ri] : ri;
  const int seg1 = p.seg_starts ? p.seg_starts[ri + 1] : ri + 1;
  int qs, qe, atype;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  // static priority for the second-dispatched half of an 8-wave WG: the
  // younger waves are the VALU-arbitration losers on every segment
  // (MI355X guide, two-waves-per-SIMD item 4)
  if (WAVES == 8 && wave >= 4) asm volatile("s_setprio 1");

  const int lo32 = lane & 31;
  const int hi = lane >> 5;

  const int n0 = nblk0 + wave * (BWD_BN * NCOL);  // this wave's k tile(s)
  const bool wave_live = n0 < ke;
  const bool skip_dq = (p.debug_ablate & 1) != 0;

  const float sl2 = HAS_SOFTCAP ? p.softcap * 1.4426950408889634f
                                : p.scale * 1.4426950408889634f;
  const float cap_pre = HAS_SOFTCAP ? p.scale / p.softcap : 0.f;
  const float log2e = 1.4426950408889634f;

  const int kh = h / p.gqa;
  const size_t k_pitch = (size_t)p.hk * D;
  const size_t q_pitch = (size_t)p.hq * D;

  // LDS: double-buffered Q/dO tiles (XOR-swizzled rows, filled by
  // global_load_lds with the swizzle on the SOURCE address — rule 21) +
  // per-wave P/dS transpose tiles. One barrier per iteration; the glds for
  // tile t+1 flies under tile t's compute and is drained by the next
  // barrier's implicit vmcnt(0).
  // ONE shared object only: a second __shared__ array makes hipcc drain
  // vmcnt(0) before every ds_read, destroying the glds pipeline (guide §5
  // ".s-level traps" (a)).
  // NBUF x 64-row Q/dO images: one barrier per 64 q rows — the doubled
  // compute phase covers the prefetch latency a 32-row phase could not.
  // MODE 0 (fused, r2-v2): BOTH the K and V tiles of every wave live in LDS
  // (2 x 64 KB) and the Q/dO ring shrinks to 32-row images (32 KB) — that
  // is the LDS budget to the byte, so lse/dpsum move from LDS to lane
  // registers + ds_bpermute. This removes the per-subtile K re-reads that
  // made the previous fused kernel fabric-bound (PMC: 172 GB/launch of
  // L2-miss fetch vs the split kernels' ~70) AND leaves ~36 registers of
  // scheduler slack below the 256 cap.
  // (the 32-row ring is only needed at D=128, where the K/V tiles take
  // 128 KB; at D=64 everything fits beside a 64-row ring)
  // Ring depth: the fused modes' K/V tiles crowd the ring out of LDS at
  // D=128 when the WG spans 256 k rows (W8, or W4x2col); the W6 fused
  // variant (192 k rows -> 96 KB of tiles) fits a 64-row ring exactly.
  constexpr int QITER =
      (V_IN_LDS && D == 128 && WAVES * NCOL >= 8) ? BWD_BM : 2 * BWD_BM;
  constexpr int NSUB = QITER / BWD_BM;
  constexpr int KVLDS = V_IN_LDS ? 2 * WAVES * NCOL * BWD_BN * ROWB : 0;
  constexpr int LSELDS = V_IN_LDS ? 0 : NBUF * 2 * QITER * 4;
  static_assert(NBUF * 2 * QITER * ROWB + LSELDS + KVLDS <= 163840,
                "LDS budget");
  __shared__ __attribute__((aligned(16))) char smem[
      NBUF * 2 * QITER * ROWB + LSELDS + KVLDS];
  auto lds_q = [&](int buf) -> __bf16* {
    return (__bf16*)(smem + (2 * buf) * QITER * ROWB);
  };
  auto lds_do = [&](int buf) -> __bf16* {
    return (__bf16*)(smem + (2 * buf + 1) * QITER * ROWB);
  };
  // lse / dpsum, staged by LDS-DMA (one dword per lane: lanes 0-31 lse,
  // 32-63 dpsum), one call per 32-row group: layout per buffer is
  // [lse g0 (32) | dps g0 (32) | lse g1 (32) | dps g1 (32)]
  auto lds_lse = [&](int buf) -> float* {
    return (float*)(smem + NBUF * 2 * QITER * ROWB + buf * 2 * QITER * 4);
  };
  // MODE 0: per-wave K and V tiles (BWD_BN rows x D, same 32-B XOR swizzle
  // as the Q/dO images so the A-fragment reads reuse the qf addressing)
  __bf16* lds_kt = (__bf16*)(smem + NBUF * 2 * QITER * ROWB + LSELDS) +
                   wave * NCOL * BWD_BN * ROWE;
  __bf16* lds_vt = (__bf16*)(smem + NBUF * 2 * QITER * ROWB + LSELDS +
                             KVLDS / 2) +
                   wave * NCOL * BWD_BN * ROWE;

  // K fragments (A-layout), loaded once per block; V fragments likewise in
  // the split dK mode — the fused mode stages V into LDS instead (register
  // budget).
  const int krow = n0 + lo32;
  const int kcl = min(krow, ke - 1);
  // MODE 0 does NOT keep K fragments live across the q loop: the lane re-reads
  // its own K row per subtile (8 b128 loads, L1/L2-resident) so the registers
  // are subtile-local — keeping them loop-live was exactly what pushed the
  // fused kernel to 437 regs / scratch spills.
  const bf16_t* kp_row = p.k + (size_t)kcl * k_pitch + (size_t)kh * D + hi * 8;
  bf16x8 kfA[V_IN_LDS ? 1 : DF], vfA[(WANT_DK && !V_IN_LDS) ? DF : 1];
  if constexpr (!V_IN_LDS) {
    const bf16_t* vp = p.v + (size_t)kcl * k_pitch + (size_t)kh * D;
#pragma unroll
    for (int dd = 0; dd < DF; ++dd) {
      kfA[dd] = *(const bf16x8*)(kp_row - hi * 8 + dd * 16 + hi * 8);
      if constexpr (WANT_DK)
        vfA[dd] = *(const bf16x8*)(vp + dd * 16 + hi * 8);
    }
  }
  constexpr int ROWS_PER_GLDS_V = 1024 / ROWB;
  if constexpr (V_IN_LDS) {
#pragma unroll
    for (int gi = 0; gi < (NCOL * BWD_BN) / ROWS_PER_GLDS_V; ++gi) {
      const int r0v = ROWS_PER_GLDS_V * gi;
      const int r = r0v + lane / (ROWB / 16);
      const int c = lane % (ROWB / 16);
      const int kr = min(n0 + r, ke - 1);
      int cs = c ^ ((r & SW32M) << 1);
      if (cs >= VSLOTS) cs = 0;
      const int csw = cs * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.k + (size_t)kr * k_pitch + (size_t)kh * D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_kt[r0v * ROWE],
          16, 0, STAGE_AUX);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.v + (size_t)kr * k_pitch + (size_t)kh * D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_vt[r0v * ROWE],
          16, 0, STAGE_AUX);
    }
  }

  // block + wave q loop bounds (recomputed per q segment)
  const int nlast = min(nblk0 + BWD_BN * NCOL * WAVES, ke) - 1;
  int q_lo = 0, q_hi = 0, wq_lo = 0, wq_hi = 0;
  auto seg_bounds = [&]() {
    q_lo = qs; q_hi = qe;
    if (atype == 1 || atype == 3) q_lo = max(q_lo, nblk0 - (ke - qe));
    if (atype == 2 || atype == 3) q_hi = min(q_hi, nlast - (ks - qs) + 1);
    wq_lo = qs; wq_hi = qe;
    if (atype == 1 || atype == 3) wq_lo = max(wq_lo, n0 - (ke - qe));
    if (atype == 2 || atype == 3)
      wq_hi = min(wq_hi, (n0 + NCOL * BWD_BN - 1) - (ks - qs) + 1);
  };

  f32x16 acc_dk[WANT_DK ? NCOL * DT : 1], acc_dv[WANT_DV ? NCOL * DT : 1];
#pragma unroll
  for (int dt = 0; dt < NCOL * DT; ++dt) {
    if constexpr (WANT_DK) acc_dk[dt] = (f32x16)(0.f);
    if constexpr (WANT_DV) acc_dv[dt] = (f32x16)(0.f);
  }

  // each wave's glds covers 4 rows (64 lanes x 16B = 1 KiB = 4 rows at D=128);
  // glds issues are dealt round-robin over the waves (wave-strided), so any
  // WAVES count works; the NBUF=3 constant-distance barrier needs a UNIFORM
  // per-wave issue count, so it requires divisibility.
  constexpr int ROWS_PER_GLDS = 1024 / ROWB;
  constexpr int GLDS_TOTAL = QITER / ROWS_PER_GLDS;
  static_assert(NBUF == 2 || GLDS_TOTAL % WAVES == 0, "uniform stage count");
  constexpr int GLDS_PER_WAVE =
      (GLDS_TOTAL % WAVES == 0) ? GLDS_TOTAL / WAVES : 0;  // 0 = non-uniform
  auto stage_glds = [&](int buf, int m0x) {
#pragma unroll
    for (int gi0 = 0; gi0 < (GLDS_TOTAL + WAVES - 1) / WAVES; ++gi0) {
      const int gi = gi0 * WAVES + wave;
      if (GLDS_TOTAL % WAVES != 0 && gi >= GLDS_TOTAL) break;
      const int r0 = ROWS_PER_GLDS * gi;
      const int r = r0 + lane / (ROWB / 16);
      const int c = lane % (ROWB / 16);
      const int qrow = min(m0x + r, qe - 1);
      int cs = c ^ ((r & SW32M) << 1);
      if (cs >= VSLOTS) cs = 0;
      const int csw = cs * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.q + (size_t)qrow * q_pitch + (size_t)h * D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_q(buf)[r0 * ROWE],
          16, 0, STAGE_AUX);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.dout + (size_t)qrow * q_pitch + (size_t)h * D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_do(buf)[r0 * ROWE],
          16, 0, STAGE_AUX);
    }
    // lse (lanes 0-31) / dpsum (lanes 32-63), one LDS-DMA per 32-row group
    // (MODE 0 has no LDS lse region: compute reads them into lane registers
    // and cross-lane-selects with ds_bpermute)
    if constexpr (!V_IN_LDS) {
#pragma unroll
      for (int sg = 0; sg < NSUB; ++sg) {
        const int qr = min(m0x + sg * BWD_BM + lo32, qe - 1);
        const float* src = (lane < 32) ? p.lse + (size_t)qr * p.hq + h
                                       : p.dpsum + (size_t)qr * p.hq + h;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)src,
            (__attribute__((address_space(3))) unsigned int*)(
                lds_lse(buf) + sg * 2 * BWD_BM),
            4, 0, STAGE_AUX);
      }
    }
  };

  // vm ops per stage call per wave (glds pairs + lse groups) — the NBUF=3
  // constant-distance barrier counts on EVERY stage issuing exactly this many
  constexpr int GOPS = 2 * GLDS_PER_WAVE + 2;
  int cur = 0;
  for (int seg = seg0; seg < seg1; ++seg) {
  qs = p.q_ranges[2 * seg];
  qe = p.q_ranges[2 * seg + 1];
  atype = p.attn_type_map ? p.attn_type_map[seg] : 0;
  if (qe <= qs) continue;  // uniform across block
  seg_bounds();
  if (q_lo >= q_hi) continue;
  cur = 0;
  stage_glds(0, q_lo);
  if constexpr (NBUF == 3) stage_glds(1, q_lo + QITER);  // rows clamp to qe-1

  for (int m0 = q_lo; m0 < q_hi; m0 += QITER) {
    // NBUF=3: wait ONLY for buf[cur] (leave the next stage in flight), then
    // prefetch two stages ahead — always, with clamped rows, so the vmcnt
    // distance stays constant. NBUF=2: full drain (prefetch had one phase).
    pipe_barrier<NBUF == 3 ? GOPS : 0>();
    if constexpr (NBUF == 3) {
      stage_glds(cur == 0 ? 2 : cur - 1, m0 + 2 * QITER);
    } else {
      if (m0 + QITER < q_hi) stage_glds(cur ^ 1, m0 + QITER);
    }
#pragma unroll
    for (int sub = 0; sub < NSUB; ++sub) {
    const int ms = m0 + sub * BWD_BM;
    if (ms >= q_hi) break;  // q_hi is block-uniform
    const __bf16* lqb = lds_q(cur) + sub * BWD_BM * ROWE;
    const __bf16* ldb = lds_do(cur) + sub * BWD_BM * ROWE;
    const float* lse_t = nullptr;
    const float* dps_t = nullptr;
    float lsedp_reg = 0.f;  // MODE0: lanes 0-31 lse, 32-63 dpsum (bpermute)
    if constexpr (!V_IN_LDS) {
      lse_t = lds_lse(cur) + sub * 2 * BWD_BM;
      dps_t = lse_t + BWD_BM;
    } else {
      const int qr = min(ms + lo32, qe - 1);
      lsedp_reg = (lane < 32) ? p.lse[(size_t)qr * p.hq + h]
                              : p.dpsum[(size_t)qr * p.hq + h];
    }
    auto lse_at = [&](int rl) -> float {
      if constexpr (!V_IN_LDS) return lse_t[rl];
      return __uint_as_float(__builtin_amdgcn_ds_bpermute(
          rl << 2, __float_as_uint(lsedp_reg)));
    };
    auto dps_at = [&](int rl) -> float {
      if constexpr (!V_IN_LDS) return dps_t[rl];
      return __uint_as_float(__builtin_amdgcn_ds_bpermute(
          (32 + rl) << 2, __float_as_uint(lsedp_reg)));
    };

    if (wave_live && ms + BWD_BM > wq_lo && ms < wq_hi) {
      // MODE 3: the wave's two 32-column tiles in sequence — each column has
      // its own C tiles and accumulator slice; Q/dO fragments and the lse /
      // dpsum selects are re-read per column (LDS/lane-local, cheap), the
      // barrier/staging cost is paid once for both.
#pragma unroll
      for (int ch = 0; ch < NCOL; ++ch) {
      const int nch = n0 + ch * BWD_BN;
      // ---- S = Q K^T ; dP = dO V^T, UN-swapped: C layout [q=crow][k=lo32],
      // so the dV/dK A-fragments come from the in-register permlane transform
      // (cframe) instead of an LDS round-trip ----
      f32x16 s = (f32x16)(0.f), dp = (f32x16)(0.f);
      if constexpr (V_IN_LDS) {
        // fused mode: K and V fragments come straight off the wave's LDS
        // tiles (staged once per block) — identical addressing to qf. The
        // two independent MFMA streams cover each other's LDS read latency.
        const char* kt_c = (const char*)(lds_kt + ch * BWD_BN * ROWE);
        const char* vt_c = (const char*)(lds_vt + ch * BWD_BN * ROWE);
#pragma unroll
        for (int dd = 0; dd < DF; ++dd) {
          const int off = swz(lo32, lo32 * ROWB + dd * 32 + hi * 16);
          bf16x8 dof = *(const bf16x8*)((const char*)ldb + off);
          bf16x8 vf = *(const bf16x8*)(vt_c + off);
          dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vf, dp, 0, 0, 0);
          bf16x8 qf = *(const bf16x8*)((const char*)lqb + off);
          bf16x8 kf = *(const bf16x8*)(kt_c + off);
          s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kf, s, 0, 0, 0);
        }
      } else {
#pragma unroll
      for (int dd = 0; dd < DF; ++dd) {
        const int off = swz(lo32, lo32 * ROWB + dd * 32 + hi * 16);
        bf16x8 qf = *(const bf16x8*)((const char*)lqb + off);
        s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kfA[dd], s, 0, 0, 0);
        if constexpr (WANT_DK) {
          bf16x8 dof = *(const bf16x8*)((const char*)ldb + off);
          dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vfA[dd], dp, 0, 0, 0);
        }
      }
      }

      const int kk = nch + lo32;  // this lane's k column
      // S is dead once P is computed, dP once dS is — reuse their registers
      // (all indices are compile-time constants after unrolling, so the
      // pointer cast stays in VGPRs; checked: ScratchSize 0)
      float* pv = (float*)&s;
      float* dsv = (float*)&dp;
      const bool interior =
          (ms + BWD_BM <= wq_hi) && (ms >= qs) && (nch + BWD_BN <= ke) &&
          !((atype == 1 || atype == 3) && (nch + BWD_BN - 1 > ms + (ke - qe))) &&
          !((atype == 2 || atype == 3) && (nch < ms + BWD_BM - 1 + (ks - qs)));
      bool all_live = interior;
      if (interior) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float lq = lse_at(crow(r, hi));
          all_live = all_live && (lq != INFINITY) && (lq != -INFINITY);
        }
      }
      if (__all(all_live) && !HAS_SOFTCAP) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int rl = crow(r, hi);
          const float pij = fast_exp2(s[r] * sl2 - lse_at(rl) * log2e);
          if constexpr (WANT_DV) pv[r] = pij;
          if constexpr (WANT_DK)
            dsv[r] = pij * (dp[r] - dps_at(rl)) * p.scale;
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrow = ms + crow(r, hi);
          const float lq = (qrow < qe) ? lse_at(crow(r, hi)) : INFINITY;
          bool ok = (qrow < wq_hi) && (qrow >= qs) && lq != INFINITY &&
                    lq != -INFINITY && kk < ke;
          if (atype == 1 || atype == 3) ok = ok && (kk - qrow <= ke - qe);
          if (atype == 2 || atype == 3) ok = ok && (kk - qrow >= ks - qs);
          float sv = s[r];
          float dscale = p.scale;
          float t;
          if (HAS_SOFTCAP) {
            const float th = tanhf(sv * cap_pre);
            t = th * sl2;
            dscale = p.scale * (1.f - th * th);
          } else {
            t = sv * sl2;
          }
          const float pij = ok ? fast_exp2(t - lq * log2e) : 0.f;
          if constexpr (WANT_DV) pv[r] = pij;
          if constexpr (WANT_DK)
            dsv[r] = pij * (dp[r] - dps_at(crow(r, hi))) * dscale;
        }
      }

      // ---- dV += P^T dO ; dK += dS^T Q ----
      // B-fragments read straight off the ROW images with ds_read_b64_tr_b16:
      // per 16-lane group, q-row bases rd*4+j supply the transpose; the 32-B
      // runs stay contiguous under the 32-B-granular swizzle. Per (dt, rd):
      //   addr = lds_base + row*ROWB + ((dcol*2) ^ ((row&7)<<5)) + (l&3)*8
      // (the XOR distributes because the swizzle bits live inside dcol*2).
      {
        const int qhalf = (lane >> 4) & 1;       // which 16-d column half
        const int jrow = (lane & 15) >> 2;       // canonical row for this lane
        const int row0 = 8 * hi + jrow;          // rd = 0 rows
        const int row1 = 8 * hi + 4 + jrow;      // rd = 1 rows
        const int q_base = (int)(unsigned long long)(
            (__attribute__((address_space(3))) const char*)lqb);
        const int do_base = (int)(unsigned long long)(
            (__attribute__((address_space(3))) const char*)ldb);
        const int lane8 = (lane & 3) * 8;
        const int sw0 = (row0 & SW32M) << 5;
        const int sw1 = (row1 & SW32M) << 5;
        const int rb0 = row0 * ROWB + lane8;
        const int rb1 = row1 * ROWB + lane8;
        // q rows 16..31 (second kk sub-tile)
        const int sw2 = ((row0 + 16) & SW32M) << 5;
        const int sw3 = ((row1 + 16) & SW32M) << 5;
        const int rb2 = (row0 + 16) * ROWB + lane8;
        const int rb3 = (row1 + 16) * ROWB + lane8;
        // software-pipelined B-frag matmul: step j+1's fragment reads issue
        // under step j's MFMA (see dq kernel note)
        auto pipe_mm = [&](int tbase, bf16x8 a0, bf16x8 a1, f32x16* acc) {
          bf16x8 bcur = tr16_frag(tbase + rb0 + ((16 * qhalf * 2) ^ sw0),
                                  tbase + rb1 + ((16 * qhalf * 2) ^ sw1));
#pragma unroll
          for (int j = 0; j < 2 * DT; ++j) {
            bf16x8 bnext = bcur;
            if (j + 1 < 2 * DT) {
              const int dcoln = (((j + 1) >> 1) * 32 + 16 * qhalf) * 2;
              if ((j + 1) & 1)
                bnext = tr16_frag(tbase + rb2 + (dcoln ^ sw2),
                                  tbase + rb3 + (dcoln ^ sw3));
              else
                bnext = tr16_frag(tbase + rb0 + (dcoln ^ sw0),
                                  tbase + rb1 + (dcoln ^ sw1));
            }
            acc[j >> 1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                (j & 1) ? a1 : a0, bcur, acc[j >> 1], 0, 0, 0);
            bcur = bnext;
          }
        };
        if constexpr (MODE == 1) {
          pipe_mm(do_base, cframe_to_afrag(pv, 0), cframe_to_afrag(pv, 1),
                  acc_dv + ch * DT);
        } else if constexpr (MODE == 2) {
          pipe_mm(q_base, cframe_to_afrag(dsv, 0), cframe_to_afrag(dsv, 1),
                  acc_dk + ch * DT);
        } else {
          // MODE 0 (fused dK+dV): INTERLEAVE the two independent MFMA
          // streams — each fragment load's LDS latency is covered by the
          // OTHER stream's in-flight MFMA, which a single serialized stream
          // cannot do (the allocator coalesces any explicit double-buffer
          // at this register pressure).
          bf16x8 pa0 = cframe_to_afrag(pv, 0);
          bf16x8 pa1 = cframe_to_afrag(pv, 1);
          bf16x8 da0 = cframe_to_afrag(dsv, 0);
          bf16x8 da1 = cframe_to_afrag(dsv, 1);
#pragma unroll
          for (int dt = 0; dt < DT; ++dt) {
            const int at = ch * DT + dt;
            const int dcol = (dt * 32 + 16 * qhalf) * 2;
            bf16x8 bdo = tr16_frag(do_base + rb0 + (dcol ^ sw0),
                                   do_base + rb1 + (dcol ^ sw1));
            acc_dv[at] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                pa0, bdo, acc_dv[at], 0, 0, 0);
            bf16x8 bq = tr16_frag(q_base + rb0 + (dcol ^ sw0),
                                  q_base + rb1 + (dcol ^ sw1));
            acc_dk[at] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                da0, bq, acc_dk[at], 0, 0, 0);
            bf16x8 bdo1 = tr16_frag(do_base + rb2 + (dcol ^ sw2),
                                    do_base + rb3 + (dcol ^ sw3));
            acc_dv[at] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                pa1, bdo1, acc_dv[at], 0, 0, 0);
            bf16x8 bq1 = tr16_frag(q_base + rb2 + (dcol ^ sw2),
                                   q_base + rb3 + (dcol ^ sw3));
            acc_dk[at] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                da1, bq1, acc_dk[at], 0, 0, 0);
          }
        }
      }
      }  // ch (column tile)
    }
    }  // sub
    cur = (NBUF == 3) ? (cur == 2 ? 0 : cur + 1) : (cur ^ 1);
  }
  __syncthreads();  // this segment's LDS reads retired before restaging
  }  // seg

  // ---- write dK/dV ----
  if (wave_live && !(p.debug_ablate & 2)) {
#pragma unroll
  for (int ch = 0; ch < NCOL; ++ch) {
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kr = n0 + ch * BWD_BN + crow(r, hi);
    if (kr >= ke) continue;
    float* dkp = p.dk + (size_t)kr * k_pitch + (size_t)kh * D;
    float* dvp = p.dv + (size_t)kr * k_pitch + (size_t)kh * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      if constexpr (WANT_DK) {
        if (acc_dk[ch * DT + dt][r] != 0.f)
          unsafeAtomicAdd(dkp + dt * 32 + lo32, acc_dk[ch * DT + dt][r]);
      }
      if constexpr (WANT_DV) {
        if (acc_dv[ch * DT + dt][r] != 0.f)
          unsafeAtomicAdd(dvp + dt * 32 + lo32, acc_dv[ch * DT + dt][r]);
      }
    }
  }
  }  // ch
  }  // wave_live store
  if constexpr (MODE == 0 || MODE == 3) break;  // single trip: kills the backedge (regs)
  }  // strided work loop
}


// ---------------- dQ pass ----------------
// q-outer: each wave owns a 32-row q tile and accumulates its dQ across the
// whole k loop IN REGISTERS, storing once at the end (atomicAdd only for the
// slice-overlap case) — eliminates the per-k-tile dq atomic storm, which the
// ablation measured at ~45% of a fused backward. K/V tiles are staged
// cooperatively per iteration (row-major swizzled for the S^T/dP^T A-frags,
// plus a transposed copy for the dQ B-frags).
template <int D, bool HAS_SOFTCAP, int WAVES, int NBUF, int NT = 0>
__global__ __launch_bounds__(64 * WAVES, WAVES == 8 ? 1 : 2)
void ffa_bwd_dq_kernel(BwdParams p) {
  constexpr int STAGE_AUX = NT ? 2 : 0;  // see dkv kernel note
  constexpr int DF = D / 16;
  constexpr int DT = D / 32;
  constexpr int ROWB = (D == 192) ? 512 : D * 2;  // padded pow2 LDS rows
  constexpr int ROWE = ROWB / 2;
  constexpr int VSLOTS = D / 8;
  constexpr int SW32M = ROWB / 32 - 1;  // 32-B-granular swizzle (tr16 reads)
  auto swz = [](int row, int byte_off) {
    return byte_off ^ ((row & SW32M) << 5);
  };
  // CU-margin strided-grid mode (see ffa_fwd.hip): margin>0 caps the grid
  // below the resident-WG count; each WG walks the flattened work space.
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
  const int ri = wz;
  const int h = p.head_major ? wx : wy;
  const int wb = p.head_major ? wy : wx;
  const int qs = p.q_ranges[2 * ri], qe = p.q_ranges[2 * ri + 1];
  const int mblk0 = qs + wb * (BWD_BM * WAVES);
  if (mblk0 >= qe) continue;
  const int seg0 = p.seg_starts ? p.seg_starts[ri] : ri;
  const int seg1 = p.seg_starts ? p.seg_starts[ri + 1] : ri + 1;
  int ks, ke, atype;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  // static priority for the second-dispatched half of an 8-wave WG: the
  // younger waves are the VALU-arbitration losers on every segment
  // (MI355X guide, two-waves-per-SIMD item 4)
  if (WAVES == 8 && wave >= 4) asm volatile("s_setprio 1");

  const int lo32 = lane & 31;
  const int hi = lane >> 5;

  const int m0 = mblk0 + wave * BWD_BM;  // this wave's q tile
  const int qrow = m0 + lo32;
  const bool qvalid = qrow < qe && m0 < qe;
  const int qcl = qvalid ? qrow : (qe - 1);
  const bool skip_dq = (p.debug_ablate & 1) != 0;

  const float sl2 = HAS_SOFTCAP ? p.softcap * 1.4426950408889634f
                                : p.scale * 1.4426950408889634f;
  const float cap_pre = HAS_SOFTCAP ? p.scale / p.softcap : 0.f;
  const float log2e = 1.4426950408889634f;

  const int kh = h / p.gqa;
  const size_t k_pitch = (size_t)p.hk * D;
  const size_t q_pitch = (size_t)p.hq * D;

  // single shared object (glds-pipeline trap, see dkv kernel).
  // 2 buffers x 64-row K/V images: one barrier per 64 k rows — the doubled
  // compute phase covers the prefetch latency a 32-row phase could not, with
  // HALF the barrier parking (PMC: SQ_WAIT_ANY 36% at 32-row iterations).
  constexpr int KITER = 2 * BWD_BN;
  static_assert(NBUF * 2 * KITER * ROWB <= 163840, "LDS budget");
  __shared__ __attribute__((aligned(16))) char smem[NBUF * 2 * KITER * ROWB];
  auto lds_k = [&](int buf) -> __bf16* {
    return (__bf16*)(smem + (2 * buf) * KITER * ROWB);
  };
  auto lds_v = [&](int buf) -> __bf16* {
    return (__bf16*)(smem + (2 * buf + 1) * KITER * ROWB);
  };

  // persistent per-wave operands: Q and dO fragments (B-layout rows)
  bf16x8 qf[DF], dof[DF];
  {
    const bf16_t* qp = p.q + (size_t)qcl * q_pitch + (size_t)h * D;
    const bf16_t* dp = p.dout + (size_t)qcl * q_pitch + (size_t)h * D;
#pragma unroll
    for (int dd = 0; dd < DF; ++dd) {
      qf[dd] = *(const bf16x8*)(qp + dd * 16 + hi * 8);
      dof[dd] = *(const bf16x8*)(dp + dd * 16 + hi * 8);
    }
  }
  const float lse_q = qvalid ? p.lse[(size_t)qrow * p.hq + h] : INFINITY;
  const float dpsum_q = qvalid ? p.dpsum[(size_t)qrow * p.hq + h] : 0.f;
  const bool row_live = qvalid && lse_q != INFINITY && lse_q != -INFINITY;

  // block + wave k-loop bounds (recomputed per k segment)
  const int mlast = min(mblk0 + BWD_BM * WAVES, qe) - 1;
  int k_lo = 0, k_hi = 0, wk_lo = 0, wk_hi = 0;
  auto seg_bounds = [&]() {
    k_lo = ks; k_hi = ke;
    if (atype == 1 || atype == 3) k_hi = min(k_hi, mlast + (ke - qe) + 1);
    if (atype == 2 || atype == 3) k_lo = max(k_lo, mblk0 + (ks - qs));
    wk_lo = ks; wk_hi = ke;
    if (atype == 1 || atype == 3)
      wk_hi = min(wk_hi, min(m0 + BWD_BM, qe) - 1 + (ke - qe) + 1);
    if (atype == 2 || atype == 3) wk_lo = max(wk_lo, m0 + (ks - qs));
  };

  f32x16 acc_dq[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) acc_dq[dt] = (f32x16)(0.f);

  constexpr int ROWS_PER_GLDS = 1024 / ROWB;
  static_assert(KITER / WAVES >= ROWS_PER_GLDS, "stage rows per wave");
  constexpr int GLDS_PER_WAVE = (KITER / WAVES) / ROWS_PER_GLDS;
  auto stage_glds = [&](int buf, int n0x) {
#pragma unroll
    for (int gi = 0; gi < GLDS_PER_WAVE; ++gi) {
      const int r0 = (KITER / WAVES) * wave + ROWS_PER_GLDS * gi;
      const int r = r0 + lane / (ROWB / 16);
      const int c = lane % (ROWB / 16);
      const int kr = min(n0x + r, ke - 1);
      int cs = c ^ ((r & SW32M) << 1);
      if (cs >= VSLOTS) cs = 0;
      const int csw = cs * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.k + (size_t)kr * k_pitch + (size_t)kh * D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_k(buf)[r0 * ROWE],
          16, 0, STAGE_AUX);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.v + (size_t)kr * k_pitch + (size_t)kh * D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_v(buf)[r0 * ROWE],
          16, 0, STAGE_AUX);
    }
  };

  constexpr int GOPS = 2 * GLDS_PER_WAVE;  // vm ops per stage per wave
  int cur = 0;
  for (int seg = seg0; seg < seg1; ++seg) {
  ks = p.k_ranges[2 * seg];
  ke = p.k_ranges[2 * seg + 1];
  atype = p.attn_type_map ? p.attn_type_map[seg] : 0;
  if (ke <= ks) continue;  // uniform across block
  seg_bounds();
  if (k_lo >= k_hi) continue;
  cur = 0;
  stage_glds(0, k_lo);
  if constexpr (NBUF == 3) stage_glds(1, k_lo + KITER);  // rows clamp to ke-1

  for (int n0 = k_lo; n0 < k_hi; n0 += KITER) {
    // NBUF=3: constant-distance barrier — wait only for buf[cur], keep the
    // next stage in flight, prefetch two ahead (clamped). NBUF=2: full drain.
    pipe_barrier<NBUF == 3 ? GOPS : 0>();
    if constexpr (NBUF == 3) {
      stage_glds(cur == 0 ? 2 : cur - 1, n0 + 2 * KITER);
    } else {
      if (n0 + KITER < k_hi) stage_glds(cur ^ 1, n0 + KITER);
    }
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
    const int ns = n0 + sub * BWD_BN;
    if (ns >= k_hi) break;  // k_hi is block-uniform
    const __bf16* lkb = lds_k(cur) + sub * BWD_BN * ROWE;
    const __bf16* lvb = lds_v(cur) + sub * BWD_BN * ROWE;

    if (wave_alive(m0, qe) && ns + BWD_BN > wk_lo && ns < wk_hi) {
      // ---- S^T = K Q^T ; dP^T = V dO^T (K/V A-frags from LDS rows) ----
      f32x16 sA = (f32x16)(0.f), dpA = (f32x16)(0.f);
#pragma unroll
      for (int dd = 0; dd < DF; ++dd) {
        const int off = swz(lo32, lo32 * ROWB + dd * 32 + hi * 16);
        bf16x8 kf = *(const bf16x8*)((const char*)lkb + off);
        bf16x8 vf = *(const bf16x8*)((const char*)lvb + off);
        sA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[dd], sA, 0, 0, 0);
        dpA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dof[dd], dpA, 0, 0, 0);
      }

      float dsv[16];
      const bool interior =
          (qrow < qe) && row_live && (ns >= ks) && (ns + BWD_BN <= ke) &&
          !((atype == 1 || atype == 3) && (ns + BWD_BN - 1 > m0 + (ke - qe))) &&
          !((atype == 2 || atype == 3) && (ns < m0 + 31 + (ks - qs)));
      if (__all(interior) && !HAS_SOFTCAP) {
        const float lsc = lse_q * log2e;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float pij = fast_exp2(sA[r] * sl2 - lsc);
          dsv[r] = pij * (dpA[r] - dpsum_q) * p.scale;
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kk = ns + crow(r, hi);
          bool ok = row_live && kk < ke && kk >= ks;
          if (atype == 1 || atype == 3) ok = ok && (kk - qrow <= ke - qe);
          if (atype == 2 || atype == 3) ok = ok && (kk - qrow >= ks - qs);
          float sv = sA[r];
          float dscale = p.scale;
          float t;
          if (HAS_SOFTCAP) {
            const float th = tanhf(sv * cap_pre);
            t = th * sl2;
            dscale = p.scale * (1.f - th * th);
          } else {
            t = sv * sl2;
          }
          const float pij = ok ? fast_exp2(t - lse_q * log2e) : 0.f;
          dsv[r] = pij * (dpA[r] - dpsum_q) * dscale;
        }
      }

      // ---- dq += dS K (B-frags from the transposed K tile) ----
      bf16x8 dsa0 = cframe_to_afrag(dsv, 0);
      bf16x8 dsa1 = cframe_to_afrag(dsv, 1);
      // B-frags (B[kk=k][j=d]) via tr16 straight off the K row image
      // (same recipe as the dkv kernel)
      {
        const int qhalf2 = (lane >> 4) & 1;
        const int jrow = (lane & 15) >> 2;
        const int row0 = 8 * hi + jrow;
        const int row1 = 8 * hi + 4 + jrow;
        const int k_base = (int)(unsigned long long)(
            (__attribute__((address_space(3))) const char*)lkb);
        const int lane8 = (lane & 3) * 8;
        const int sw0 = (row0 & SW32M) << 5;
        const int sw1 = (row1 & SW32M) << 5;
        const int rb0 = row0 * ROWB + lane8;
        const int rb1 = row1 * ROWB + lane8;
        const int sw2 = ((row0 + 16) & SW32M) << 5;
        const int sw3 = ((row1 + 16) & SW32M) << 5;
        const int rb2 = (row0 + 16) * ROWB + lane8;
        const int rb3 = (row1 + 16) * ROWB + lane8;
        // software-pipelined: step j+1's fragment reads issue under step j's
        // MFMA so the ~50-cycle LDS latency hides under the matrix op (the
        // serialized read->wait->mfma form parks the wave per fragment)
        bf16x8 bcur = tr16_frag(k_base + rb0 + ((16 * qhalf2 * 2) ^ sw0),
                                k_base + rb1 + ((16 * qhalf2 * 2) ^ sw1));
#pragma unroll
        for (int j = 0; j < 2 * DT; ++j) {
          bf16x8 bnext = bcur;
          if (j + 1 < 2 * DT) {
            const int dcoln = (((j + 1) >> 1) * 32 + 16 * qhalf2) * 2;
            if ((j + 1) & 1)
              bnext = tr16_frag(k_base + rb2 + (dcoln ^ sw2),
                                k_base + rb3 + (dcoln ^ sw3));
            else
              bnext = tr16_frag(k_base + rb0 + (dcoln ^ sw0),
                                k_base + rb1 + (dcoln ^ sw1));
          }
          acc_dq[j >> 1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              (j & 1) ? dsa1 : dsa0, bcur, acc_dq[j >> 1], 0, 0, 0);
          bcur = bnext;
        }
      }
    }
    }  // sub
    cur = (NBUF == 3) ? (cur == 2 ? 0 : cur + 1) : (cur ^ 1);
  }
  __syncthreads();  // this segment's LDS reads retired before restaging
  }  // seg

  // ---- store dq once (atomicAdd: q_ranges of different slices may overlap) ----
  if (!skip_dq) {
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qr = m0 + crow(r, hi);
    if (qr >= qe) continue;
    float* dst = p.dq + (size_t)qr * q_pitch + (size_t)h * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      const float val = acc_dq[dt][r];
      if (val != 0.f) unsafeAtomicAdd(dst + dt * 32 + lo32, val);
    }
  }
  }  // !skip_dq
  }  // strided work loop
}


// ---------------- dQ pass, 64-row q tiles (r2) ----------------
// Each wave owns a 64-row q tile as two interleaved 32-row halves at ONE
// wave/SIMD (512-reg budget). Why this beats the 32-row/2-wave shape:
// every K/V fragment read (b128 A-frags and tr16 B-frags) is SHARED by the
// two halves' MFMAs — memory ops per MFMA halve — and the two halves are
// independent MFMA streams, so each read's LDS latency hides under the
// other half's matrix op without a partner wave.
template <int D, bool HAS_SOFTCAP, int NBUF, int NT = 0>
__global__ __launch_bounds__(256, 1)
void ffa_bwd_dq64_kernel(BwdParams p) {
  constexpr int STAGE_AUX = NT ? 2 : 0;
  constexpr int DF = D / 16;
  constexpr int DT = D / 32;
  constexpr int ROWB = D * 2;
  constexpr int SW32M = ROWB / 32 - 1;
  constexpr int WAVES = 4;
  constexpr int TQ = 64;  // q rows per wave
  auto swz = [](int row, int byte_off) {
    return byte_off ^ ((row & SW32M) << 5);
  };
  const int ri = blockIdx.z;
  const int h = p.head_major ? blockIdx.x : blockIdx.y;
  const int wb = p.head_major ? blockIdx.y : blockIdx.x;
  const int qs = p.q_ranges[2 * ri], qe = p.q_ranges[2 * ri + 1];
  const int mblk0 = qs + wb * (TQ * WAVES);
  if (mblk0 >= qe) return;
  const int seg0 = p.seg_starts ? p.seg_starts[ri] : ri;
  const int seg1 = p.seg_starts ? p.seg_starts[ri + 1] : ri + 1;
  int ks, ke, atype;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int lo32 = lane & 31;
  const int hi = lane >> 5;

  const int m0 = mblk0 + wave * TQ;  // this wave's 64-row q tile
  const bool skip_dq = (p.debug_ablate & 1) != 0;

  const float sl2 = HAS_SOFTCAP ? p.softcap * 1.4426950408889634f
                                : p.scale * 1.4426950408889634f;
  const float cap_pre = HAS_SOFTCAP ? p.scale / p.softcap : 0.f;
  const float log2e = 1.4426950408889634f;

  const int kh = h / p.gqa;
  const size_t k_pitch = (size_t)p.hk * D;
  const size_t q_pitch = (size_t)p.hq * D;

  constexpr int KITER = 2 * BWD_BN;
  static_assert(NBUF * 2 * KITER * D * 2 <= 163840, "LDS budget");
  __shared__ __attribute__((aligned(16))) char smem[NBUF * 2 * KITER * D * 2];
  auto lds_k = [&](int buf) -> __bf16* {
    return (__bf16*)(smem + (2 * buf) * KITER * D * 2);
  };
  auto lds_v = [&](int buf) -> __bf16* {
    return (__bf16*)(smem + (2 * buf + 1) * KITER * D * 2);
  };

  // persistent per-wave operands, both halves
  bf16x8 qf[2][DF], dof[2][DF];
  float lse_q[2], dpsum_q[2];
  bool row_live[2], qvalid[2];
  int qrow_h[2];
#pragma unroll
  for (int hh = 0; hh < 2; ++hh) {
    const int qrow = m0 + 32 * hh + lo32;
    qrow_h[hh] = qrow;
    qvalid[hh] = qrow < qe;
    const int qcl = qvalid[hh] ? qrow : (qe - 1);
    const bf16_t* qp = p.q + (size_t)qcl * q_pitch + (size_t)h * D;
    const bf16_t* dp = p.dout + (size_t)qcl * q_pitch + (size_t)h * D;
#pragma unroll
    for (int dd = 0; dd < DF; ++dd) {
      qf[hh][dd] = *(const bf16x8*)(qp + dd * 16 + hi * 8);
      dof[hh][dd] = *(const bf16x8*)(dp + dd * 16 + hi * 8);
    }
    lse_q[hh] = qvalid[hh] ? p.lse[(size_t)qrow * p.hq + h] : INFINITY;
    dpsum_q[hh] = qvalid[hh] ? p.dpsum[(size_t)qrow * p.hq + h] : 0.f;
    row_live[hh] = qvalid[hh] && lse_q[hh] != INFINITY &&
                   lse_q[hh] != -INFINITY;
  }

  const int mlast = min(mblk0 + TQ * WAVES, qe) - 1;
  int k_lo = 0, k_hi = 0, wk_lo = 0, wk_hi = 0;
  auto seg_bounds = [&]() {
    k_lo = ks; k_hi = ke;
    if (atype == 1 || atype == 3) k_hi = min(k_hi, mlast + (ke - qe) + 1);
    if (atype == 2 || atype == 3) k_lo = max(k_lo, mblk0 + (ks - qs));
    wk_lo = ks; wk_hi = ke;
    if (atype == 1 || atype == 3)
      wk_hi = min(wk_hi, min(m0 + TQ, qe) - 1 + (ke - qe) + 1);
    if (atype == 2 || atype == 3) wk_lo = max(wk_lo, m0 + (ks - qs));
  };

  f32x16 acc_dq[2][DT];
#pragma unroll
  for (int hh = 0; hh < 2; ++hh)
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) acc_dq[hh][dt] = (f32x16)(0.f);

  constexpr int ROWS_PER_GLDS = 1024 / ROWB;
  constexpr int GLDS_PER_WAVE = (KITER / WAVES) / ROWS_PER_GLDS;
  auto stage_glds = [&](int buf, int n0x) {
#pragma unroll
    for (int gi = 0; gi < GLDS_PER_WAVE; ++gi) {
      const int r0 = (KITER / WAVES) * wave + ROWS_PER_GLDS * gi;
      const int r = r0 + lane / (ROWB / 16);
      const int c = lane % (ROWB / 16);
      const int kr = min(n0x + r, ke - 1);
      const int csw = (c ^ ((r & SW32M) << 1)) * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.k + (size_t)kr * k_pitch + (size_t)kh * D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_k(buf)[r0 * D],
          16, 0, STAGE_AUX);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.v + (size_t)kr * k_pitch + (size_t)kh * D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_v(buf)[r0 * D],
          16, 0, STAGE_AUX);
    }
  };

  constexpr int GOPS = 2 * GLDS_PER_WAVE;
  int cur = 0;
  for (int seg = seg0; seg < seg1; ++seg) {
  ks = p.k_ranges[2 * seg];
  ke = p.k_ranges[2 * seg + 1];
  atype = p.attn_type_map ? p.attn_type_map[seg] : 0;
  if (ke <= ks) continue;
  seg_bounds();
  if (k_lo >= k_hi) continue;
  cur = 0;
  stage_glds(0, k_lo);
  if constexpr (NBUF == 3) stage_glds(1, k_lo + KITER);

  for (int n0 = k_lo; n0 < k_hi; n0 += KITER) {
    pipe_barrier<NBUF == 3 ? GOPS : 0>();
    if constexpr (NBUF == 3) {
      stage_glds(cur == 0 ? 2 : cur - 1, n0 + 2 * KITER);
    } else {
      if (n0 + KITER < k_hi) stage_glds(cur ^ 1, n0 + KITER);
    }
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
    const int ns = n0 + sub * BWD_BN;
    if (ns >= k_hi) break;
    const __bf16* lkb = lds_k(cur) + sub * BWD_BN * D;
    const __bf16* lvb = lds_v(cur) + sub * BWD_BN * D;

    if (m0 < qe && ns + BWD_BN > wk_lo && ns < wk_hi) {
      // ---- S^T / dP^T for BOTH halves off ONE kf/vf read pair ----
      f32x16 sA0 = (f32x16)(0.f), dpA0 = (f32x16)(0.f);
      f32x16 sA1 = (f32x16)(0.f), dpA1 = (f32x16)(0.f);
#pragma unroll
      for (int dd = 0; dd < DF; ++dd) {
        const int off = swz(lo32, lo32 * ROWB + dd * 32 + hi * 16);
        bf16x8 kf = *(const bf16x8*)((const char*)lkb + off);
        bf16x8 vf = *(const bf16x8*)((const char*)lvb + off);
        sA0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[0][dd], sA0, 0, 0, 0);
        dpA0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dof[0][dd], dpA0, 0, 0, 0);
        sA1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[1][dd], sA1, 0, 0, 0);
        dpA1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dof[1][dd], dpA1, 0, 0, 0);
      }

      // ---- softmax recompute per half; `scale` folded into the FINAL dq
      // store (dq is linear in dS), not the per-element dsv ----
      float dsv[2][16];
#pragma unroll
      for (int hh = 0; hh < 2; ++hh) {
        const f32x16& sA = hh ? sA1 : sA0;
        const f32x16& dpA = hh ? dpA1 : dpA0;
        const int m0h = m0 + 32 * hh;
        const bool interior =
            (qrow_h[hh] < qe) && row_live[hh] && (ns >= ks) &&
            (ns + BWD_BN <= ke) &&
            !((atype == 1 || atype == 3) &&
              (ns + BWD_BN - 1 > m0h + (ke - qe))) &&
            !((atype == 2 || atype == 3) && (ns < m0h + 31 + (ks - qs)));
        if (__all(interior) && !HAS_SOFTCAP) {
          const float lsc = lse_q[hh] * log2e;
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float pij = fast_exp2(sA[r] * sl2 - lsc);
            dsv[hh][r] = pij * (dpA[r] - dpsum_q[hh]);
          }
        } else {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kk = ns + crow(r, hi);
            bool ok = row_live[hh] && kk < ke && kk >= ks;
            if (atype == 1 || atype == 3)
              ok = ok && (kk - qrow_h[hh] <= ke - qe);
            if (atype == 2 || atype == 3)
              ok = ok && (kk - qrow_h[hh] >= ks - qs);
            float sv = sA[r];
            float drel = 1.f;
            float t;
            if (HAS_SOFTCAP) {
              const float th = tanhf(sv * cap_pre);
              t = th * sl2;
              drel = 1.f - th * th;
            } else {
              t = sv * sl2;
            }
            const float pij = ok ? fast_exp2(t - lse_q[hh] * log2e) : 0.f;
            dsv[hh][r] = pij * (dpA[r] - dpsum_q[hh]) * drel;
          }
        }
      }

      // ---- dq += dS K: ONE tr16 B-frag feeds BOTH halves' MFMAs ----
      bf16x8 a00 = cframe_to_afrag(dsv[0], 0);
      bf16x8 a01 = cframe_to_afrag(dsv[0], 1);
      bf16x8 a10 = cframe_to_afrag(dsv[1], 0);
      bf16x8 a11 = cframe_to_afrag(dsv[1], 1);
      {
        const int qhalf2 = (lane >> 4) & 1;
        const int jrow = (lane & 15) >> 2;
        const int row0 = 8 * hi + jrow;
        const int row1 = 8 * hi + 4 + jrow;
        const int k_base = (int)(unsigned long long)(
            (__attribute__((address_space(3))) const char*)lkb);
        const int lane8 = (lane & 3) * 8;
        const int sw0 = (row0 & SW32M) << 5;
        const int sw1 = (row1 & SW32M) << 5;
        const int rb0 = row0 * ROWB + lane8;
        const int rb1 = row1 * ROWB + lane8;
        const int sw2 = ((row0 + 16) & SW32M) << 5;
        const int sw3 = ((row1 + 16) & SW32M) << 5;
        const int rb2 = (row0 + 16) * ROWB + lane8;
        const int rb3 = (row1 + 16) * ROWB + lane8;
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) {
          const int dcol = (dt * 32 + 16 * qhalf2) * 2;
          bf16x8 b0 = tr16_frag(k_base + rb0 + (dcol ^ sw0),
                                k_base + rb1 + (dcol ^ sw1));
          acc_dq[0][dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a00, b0, acc_dq[0][dt], 0, 0, 0);
          acc_dq[1][dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a10, b0, acc_dq[1][dt], 0, 0, 0);
          bf16x8 b1 = tr16_frag(k_base + rb2 + (dcol ^ sw2),
                                k_base + rb3 + (dcol ^ sw3));
          acc_dq[0][dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a01, b1, acc_dq[0][dt], 0, 0, 0);
          acc_dq[1][dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a11, b1, acc_dq[1][dt], 0, 0, 0);
        }
      }
    }
    }  // sub
    cur = (NBUF == 3) ? (cur == 2 ? 0 : cur + 1) : (cur ^ 1);
  }
  __syncthreads();
  }  // seg

  if (skip_dq) return;
#pragma unroll
  for (int hh = 0; hh < 2; ++hh) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qr = m0 + 32 * hh + crow(r, hi);
      if (qr >= qe) continue;
      float* dst = p.dq + (size_t)qr * q_pitch + (size_t)h * D;
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        const float val = acc_dq[hh][dt][r] * p.scale;
        if (val != 0.f) unsafeAtomicAdd(dst + dt * 32 + lo32, val);
      }
    }
  }
}

// ---------------- launchers ----------------
extern "C" int magi_ffa_bwd_preprocess(const magi_ffa_bwd_args* a) {
  if (!a || !a->dout || !a->out || !a->dpsum) return -1;
  BwdParams p{};
  p.dout = (const bf16_t*)a->dout;
  p.out = a->out;
  p.dpsum = a->dpsum;
  const long long n_rows = a->total_q * a->hq;
  if (n_rows == 0) return 0;
  dim3 grid((unsigned)((n_rows + 3) / 4)), block(256);
  hipStream_t s = (hipStream_t)a->stream;
  if (a->out_is_fp32)
    hipLaunchKernelGGL((bwd_preprocess_kernel<true>), grid, block, 0, s, p,
                       a->d, n_rows);
  else
    hipLaunchKernelGGL((bwd_preprocess_kernel<false>), grid, block, 0, s, p,
                       a->d, n_rows);
  return (int)hipGetLastError();
}

static int fill_bwd_params(const magi_ffa_bwd_args* a, BwdParams* p) {
  if (!a || !a->dout || !a->q || !a->k || !a->v || !a->lse) return -1;
  if (a->d != 64 && a->d != 128 && a->d != 192) return -2;
  if (a->hq % a->hk != 0) return -3;
  if (a->n_ranges <= 0) return 1;  // caller treats >0 as "nothing to do"
  p->dout = (const bf16_t*)a->dout;
  p->q = (const bf16_t*)a->q;
  p->k = (const bf16_t*)a->k;
  p->v = (const bf16_t*)a->v;
  p->out = a->out;
  p->lse = a->lse;
  p->dq = a->dq;
  p->dk = a->dk;
  p->dv = a->dv;
  p->dpsum = a->dpsum;
  p->q_ranges = a->q_ranges;
  p->k_ranges = a->k_ranges;
  p->attn_type_map = a->attn_type_map;
  p->hq = a->hq;
  p->hk = a->hk;
  p->gqa = a->hq / a->hk;
  p->scale = a->softmax_scale;
  p->softcap = a->softcap;
  p->total_q = a->total_q;
  p->total_k = a->total_k;
  p->seg_starts = a->seg_starts;
  { const char* e = getenv("MAGI_BWD_ABLATE"); p->debug_ablate = e ? atoi(e) : 0; }
  // head-major pays when one head's K+V (bf16) fits a 4 MB XCD L2
  // late-r2 A/B (profiles/r2_headmajor_ab.md): dq wins head-major up to
  // ~16 MB of per-head K+V (1.20x at 16k) and loses beyond (0.93x at 64k,
  // 0.91x at 128k) — same shape as the fwd kernel; the dkv launcher
  // OVERRIDES this to always-head-major below.
  p->head_major = ((long long)a->total_k * a->d * 4 <= (16 << 20)) ? 1 : 0;
  { const char* e = getenv("MAGI_BWD_HEAD_MAJOR"); if (e) p->head_major = atoi(e); }
  // deterministic-mode GQA split: cu_margin's top bits carry (mult<<8)|off;
  // 0 = all heads in one launch (default)
  p->head_mult = 1;
  p->head_off = 0;
  p->n_heads_launch = a->hq;
  if (a->cu_margin & 0xFFFF0000) {
    p->head_mult = (a->cu_margin >> 24) & 0xFF;
    p->head_off = (a->cu_margin >> 16) & 0xFF;
    p->n_heads_launch = a->hq / p->head_mult;
  }
  return 0;
}

extern "C" int magi_ffa_bwd_dq(const magi_ffa_bwd_args* a) {
  BwdParams p;
  int rc = fill_bwd_params(a, &p);
  if (rc) return rc > 0 ? 0 : rc;

  // 8 waves share one staged K/V image (one 512-thread WG/CU, 2 waves/SIMD)
  // for LONG ranges: halves staging+barrier cost per MFMA (measured
  // 67->63 ms at 64k). Short ranges keep 4 waves — the 256-row block
  // windows over-iterate masked edges on 2k varlen docs.
  // W8 runs the 3-slot LDS ring (constant-distance vmcnt barrier); W4 cannot
  // (3 buffers exceed the 80 KB/WG budget of 2 blocks/CU) and keeps 2.
  // dq's q-outer blocks DO over-iterate causal k windows at W8 (r2 A/B on
  // 2k varlen: 0.79 -> 1.61 ms), so it keeps the long-range threshold
  int dq_bigseq = 8192;
  { const char* e = getenv("MAGI_BWD_DQ_BIGSEQ"); if (e) dq_bigseq = atoi(e); }
  // D=192 W8 builds spill at the 256-reg cap: stay at 4 waves there
  const int dqw = (a->d != 192 && a->max_seqlen_k >= dq_bigseq) ? 8 : 4;
  int nbuf = 3;
  { const char* e = getenv("MAGI_BWD_NBUF"); if (e) nbuf = atoi(e); }
  int nt = 0;
  { const char* e = getenv("MAGI_STAGE_NT"); if (e) nt = atoi(e); }
  // measured 61.1 vs 54.8 ms at 64k: without a partner wave the exposed
  // LDS/barrier latencies outweigh the halved memory ops — keep 2 waves/SIMD
  int dq64 = 0;
  { const char* e = getenv("MAGI_BWD_DQ64"); if (e) dq64 = atoi(e); }
  if (a->n_ranges > 65535) return -5;
  hipStream_t s = (hipStream_t)a->stream;
  const bool sc = a->softcap > 0.f;
  if (dqw == 8 && dq64 && a->d != 192) {
    // 64-row q tiles, 4 waves, 1 wave/SIMD (same 256-row block span)
    const int qblocks64 = (int)((a->total_q + 255) / 256);
    dim3 grid64 = p.head_major
                      ? dim3(a->hq, qblocks64, (unsigned)a->n_ranges)
                      : dim3(qblocks64, a->hq, (unsigned)a->n_ranges);
    p.work_nx = grid64.x; p.work_ny = grid64.y; p.work_nz = grid64.z;
    dim3 block64(256);
#define LAUNCH_DQ64(DD, SC) \
  hipLaunchKernelGGL((ffa_bwd_dq64_kernel<DD, SC, 3>), grid64, block64, 0, s, p)
    if (a->d == 64) { if (sc) LAUNCH_DQ64(64, true); else LAUNCH_DQ64(64, false); }
    else            { if (sc) LAUNCH_DQ64(128, true); else LAUNCH_DQ64(128, false); }
#undef LAUNCH_DQ64
    return (int)hipGetLastError();
  }
  const int qspan = BWD_BM * dqw;
  const int qblocks = (int)((a->total_q + qspan - 1) / qspan);
  dim3 grid_q = p.head_major ? dim3(a->hq, qblocks, (unsigned)a->n_ranges)
                             : dim3(qblocks, a->hq, (unsigned)a->n_ranges);
  p.work_nx = grid_q.x; p.work_ny = grid_q.y; p.work_nz = grid_q.z;
  const int margin = a->cu_margin & 0xFFFF;
  if (margin > 0) {  // CU-margin: cap resident WGs (see ffa_fwd.hip)
    const long long total = (long long)grid_q.x * grid_q.y * grid_q.z;
    const int per_cu = (dqw == 4) ? 2 : 1;
    long long cap = (long long)(256 - margin) * per_cu;
    if (cap < 1) cap = 1;
    if (cap > total) cap = total;
    grid_q = dim3((unsigned)cap, 1, 1);
  }
  dim3 block(64 * dqw);
#define LAUNCH_DQ(DD, SC, WW, NB, NTV) \
  hipLaunchKernelGGL((ffa_bwd_dq_kernel<DD, SC, WW, NB, NTV>), grid_q, block, \
                     0, s, p)
#define PICK_DQ_NT(DD, NTV) \
  do { \
    if (dqw == 8) { \
      bool done3 = false; \
      if constexpr (DD != 192) { /* a 3-ring of 512-B rows exceeds LDS */ \
        if (nbuf == 3) { \
          if (sc) LAUNCH_DQ(DD, true, 8, 3, NTV); else LAUNCH_DQ(DD, false, 8, 3, NTV); \
          done3 = true; \
        } \
      } \
      if (!done3) { \
        if (sc) LAUNCH_DQ(DD, true, 8, 2, NTV); else LAUNCH_DQ(DD, false, 8, 2, NTV); \
      } \
    } else { \
      if (sc) LAUNCH_DQ(DD, true, 4, 2, NTV); else LAUNCH_DQ(DD, false, 4, 2, NTV); \
    } \
  } while (0)
#define PICK_DQ(DD) do { if (nt) PICK_DQ_NT(DD, 1); else PICK_DQ_NT(DD, 0); } while (0)
  if (a->d == 64) PICK_DQ(64);
  else if (a->d == 192) PICK_DQ(192);
  else PICK_DQ(128);
#undef PICK_DQ
#undef PICK_DQ_NT
#undef LAUNCH_DQ
  return (int)hipGetLastError();
}

template <int MODE>
static int launch_bwd_dkv(const magi_ffa_bwd_args* a) {
  BwdParams p;
  int rc = fill_bwd_params(a, &p);
  if (rc) return rc > 0 ? 0 : rc;
  // dkv passes win head-major at EVERY measured size (their dk/dv atomics
  // cluster per head in one XCD's L2): 1.54x at 16k, 1.10x at 64k, 1.07x
  // at 128k (profiles/r2_headmajor_ab.md) — unconditional unless A/B'd
  if (!getenv("MAGI_BWD_HEAD_MAJOR")) p.head_major = 1;
  // All dkv modes run 8 waves per WG (one 512-thread WG/CU, one shared
  // staged Q/dO image) down to 1k ranges: these kernels' blocks span the
  // K dim, so the wider workgroup does NOT over-iterate causal q windows,
  // and halving the staging streams is a 50-60% win even on 2k varlen docs
  // (r2 A/B: dv 0.71->0.48 ms, dk 0.90->0.56, fused 1.37->0.92). The fused
  // mode keeps NBUF=2 (its V tiles + a 3-ring exceed the 160 KB LDS); the
  // split W8 modes default to the 3-slot ring.
  int bigseq = 1024;
  { const char* e = getenv("MAGI_BWD_BIGSEQ"); if (e) bigseq = atoi(e); }
  const bool big = a->d != 192 && a->max_seqlen_k >= bigseq;
  // W8 for ALL modes on long ranges: one 512-thread WG/CU halves the
  // staging streams (PMC r2: dV at W4 fetched 174 GB/launch vs dK-W8's 71)
  const int W = big ? 8 : 4;
  // MODE 3 "fat" fused: 4 waves x 2 column tiles at 1 wave/SIMD (same
  // 256-row span/grid as MODE 0 at W8) — A/B gate, see kernel note
  int fat = 0;
  { const char* e = getenv("MAGI_BWD_FAT"); if (e) fat = atoi(e); }
  const bool use_fat = (MODE == 0) && big && fat != 0;
  // W6 fused: 192-row k span, 64-row Q/dO ring (2x the MFMAs per barrier of
  // the W8 32-row ring), 1.5 waves/SIMD — A/B gate
  int w6 = 0;
  { const char* e = getenv("MAGI_BWD_W6"); if (e) w6 = atoi(e); }
  const bool use_w6 = (MODE == 0) && big && !use_fat && w6 != 0;
  int nbuf = (MODE == 0 || W == 4) ? 2 : 3;
  { const char* e = getenv("MAGI_BWD_NBUF");
    if (e && nbuf == 3) nbuf = atoi(e); }
  int nt = 0;
  { const char* e = getenv("MAGI_STAGE_NT"); if (e) nt = atoi(e); }
  const int span =
      use_fat ? (2 * 4 * BWD_BN) : (use_w6 ? 6 * BWD_BN : BWD_BN * W);
  const int nblocks = (a->max_seqlen_k + span - 1) / span;
  if (a->n_ranges > 65535) return -5;
  dim3 grid_kv = p.head_major
                     ? dim3(p.n_heads_launch, nblocks, (unsigned)a->n_ranges)
                     : dim3(nblocks, p.n_heads_launch, (unsigned)a->n_ranges);
  p.work_nx = grid_kv.x; p.work_ny = grid_kv.y; p.work_nz = grid_kv.z;
  const int margin = (MODE == 0) ? 0 : (a->cu_margin & 0xFFFF);
  if (margin > 0) {  // CU-margin: cap resident WGs (see ffa_fwd.hip)
    const long long total = (long long)grid_kv.x * grid_kv.y * grid_kv.z;
    const int per_cu = (W == 4 && MODE != 0) ? 2 : 1;
    long long cap = (long long)(256 - margin) * per_cu;
    if (cap < 1) cap = 1;
    if (cap > total) cap = total;
    grid_kv = dim3((unsigned)cap, 1, 1);
  }
  dim3 block(use_fat ? 256 : (use_w6 ? 384 : 64 * W));
  hipStream_t s = (hipStream_t)a->stream;
  const bool sc = a->softcap > 0.f;
#define LAUNCH_DKV(DD, SC, WW, NB, NTV) \
  hipLaunchKernelGGL((ffa_bwd_dkv_kernel<DD, SC, MODE, WW, NB, NTV>), grid_kv, \
                     block, 0, s, p)
#define LAUNCH_FAT(DD, SC, NTV) \
  hipLaunchKernelGGL((ffa_bwd_dkv_kernel<DD, SC, 3, 4, 2, NTV>), grid_kv, \
                     block, 0, s, p)
#define LAUNCH_W6(DD, SC, NTV) \
  hipLaunchKernelGGL((ffa_bwd_dkv_kernel<DD, SC, 0, 6, 2, NTV>), grid_kv, \
                     block, 0, s, p)
#define PICK_DKV_NT(DD, NTV) \
  do { \
    if (use_fat) { \
      if constexpr (MODE == 0 && DD != 192) { \
        if (sc) LAUNCH_FAT(DD, true, NTV); else LAUNCH_FAT(DD, false, NTV); \
        break; \
      } \
    } \
    if (use_w6) { \
      if constexpr (MODE == 0 && DD != 192) { \
        if (sc) LAUNCH_W6(DD, true, NTV); else LAUNCH_W6(DD, false, NTV); \
        break; \
      } \
    } \
    if (W == 8) { \
      bool done = false; \
      if constexpr (MODE != 0 && DD != 192) { /* 3-ring LDS limits */ \
        if (nbuf == 3) { \
          if (sc) LAUNCH_DKV(DD, true, 8, 3, NTV); else LAUNCH_DKV(DD, false, 8, 3, NTV); \
          done = true; \
        } \
      } \
      if (!done) { \
        if (sc) LAUNCH_DKV(DD, true, 8, 2, NTV); else LAUNCH_DKV(DD, false, 8, 2, NTV); \
      } \
    } else { \
      if (sc) LAUNCH_DKV(DD, true, 4, 2, NTV); else LAUNCH_DKV(DD, false, 4, 2, NTV); \
    } \
  } while (0)
#define PICK_DKV(DD) do { if (nt) PICK_DKV_NT(DD, 1); else PICK_DKV_NT(DD, 0); } while (0)
  if (a->d == 64) PICK_DKV(64);
  else if (a->d == 192) {
    if constexpr (MODE != 0) PICK_DKV(192);
    else return -6;  // fused + D=192 exceeds LDS; host routes to the split
  } else PICK_DKV(128);
#undef PICK_DKV
#undef PICK_DKV_NT
#undef LAUNCH_W6
#undef LAUNCH_FAT
#undef LAUNCH_DKV
  return (int)hipGetLastError();
}

extern "C" int magi_ffa_bwd_dkv(const magi_ffa_bwd_args* a) {
  return launch_bwd_dkv<0>(a);
}

extern "C" int magi_ffa_bwd_dv(const magi_ffa_bwd_args* a) {
  return launch_bwd_dkv<1>(a);
}

extern "C" int magi_ffa_bwd_dk(const magi_ffa_bwd_args* a) {
  return launch_bwd_dkv<2>(a);
}

extern "C" int magi_ffa_bwd(const magi_ffa_bwd_args* a) {
  // sequential convenience form: dq pass then dkv pass on one stream
  int rc = magi_ffa_bwd_dq(a);
  if (rc) return rc;
  return magi_ffa_bwd_dkv(a);
}
