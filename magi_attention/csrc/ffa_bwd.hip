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

DEV_INLINE int crow(int r, int hi) { return (r & 3) + 8 * (r >> 2) + 4 * hi; }

DEV_INLINE unsigned pack_bf16_pair(float lo, float hi) {
  union { __bf16 b; unsigned short u; } a, b;
  a.b = (__bf16)lo;
  b.b = (__bf16)hi;
  return ((unsigned)b.u << 16) | a.u;
}

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
template <int D, bool HAS_SOFTCAP>
__global__ __launch_bounds__(256, 1) void ffa_bwd_kernel(BwdParams p) {
  constexpr int DF = D / 16;
  constexpr int DT = D / 32;
  constexpr int ROWB = D * 2;  // bytes per LDS tile row
  // XOR swizzle within one LDS row: spreads a b128 lane group over the row's
  // 16-B slots (T2/G4); mask keeps the XOR inside the row for D=64 too.
  constexpr int SWZM = ROWB / 16 - 1;
  auto swz = [](int row, int byte_off) {
    return byte_off ^ ((row & SWZM) << 4);
  };
  // blockIdx.x = HEAD: the dispatcher places block b on XCD b%8, so every
  // block of one head shares one XCD/L2 — dq/dk/dv atomics stay XCD-local and
  // K/V reads of a head hit a single L2 (T1 XCD-affinity via grid layout).
  const int ri = blockIdx.z;
  const int h = blockIdx.x;
  const int ks = p.k_ranges[2 * ri], ke = p.k_ranges[2 * ri + 1];
  const int nblk0 = ks + blockIdx.y * (BWD_BN * BWD_WAVES);
  if (nblk0 >= ke) return;
  const int qs = p.q_ranges[2 * ri], qe = p.q_ranges[2 * ri + 1];
  if (qe <= qs) return;
  const int atype = p.attn_type_map ? p.attn_type_map[ri] : 0;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int lo32 = lane & 31;
  const int hi = lane >> 5;

  const int n0 = nblk0 + wave * BWD_BN;  // this wave's k tile
  const bool wave_live = n0 < ke;
  const bool skip_dq = (p.debug_ablate & 1) != 0;

  const float sl2 = HAS_SOFTCAP ? p.softcap * 1.4426950408889634f
                                : p.scale * 1.4426950408889634f;
  const float cap_pre = HAS_SOFTCAP ? p.scale / p.softcap : 0.f;
  const float log2e = 1.4426950408889634f;

  const int kh = h / p.gqa;
  const size_t k_pitch = (size_t)p.hk * D;
  const size_t q_pitch = (size_t)p.hq * D;

  // LDS: Q/dO tiles (XOR-swizzled rows) + their transposed copies (padded to
  // 40-elem rows so b128 reads stay 16-B aligned and bank-spread) + per-wave
  // P/dS transpose tiles + the block-combined dQ tile (4 waves' contributions
  // summed in LDS, flushed by ONE global-atomic pass -> 4x fewer HBM atomics;
  // ablation showed dq atomics were 45% of bwd time).
  __shared__ __bf16 lds_q[BWD_BM * D];
  __shared__ __bf16 lds_do[BWD_BM * D];
  __shared__ __bf16 lds_p[BWD_WAVES][32][34];

  // K/V fragments (A-layout) + K B-fragments, loaded once per block
  const int krow = n0 + lo32;
  const int kcl = min(krow, ke - 1);
  bf16x8 kfA[DF], vfA[DF];
  {
    const bf16_t* kp = p.k + (size_t)kcl * k_pitch + (size_t)kh * D;
    const bf16_t* vp = p.v + (size_t)kcl * k_pitch + (size_t)kh * D;
#pragma unroll
    for (int dd = 0; dd < DF; ++dd) {
      kfA[dd] = *(const bf16x8*)(kp + dd * 16 + hi * 8);
      vfA[dd] = *(const bf16x8*)(vp + dd * 16 + hi * 8);
    }
  }
  union Bf {
    unsigned short u[8];
    bf16x8 v;
  };
  bf16x8 kfB[DT][2];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt)
#pragma unroll
    for (int tt = 0; tt < 2; ++tt) {
      Bf b;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int row = min(n0 + 16 * tt + 8 * hi + e, ke - 1);
        b.u[e] = *(const unsigned short*)(p.k + (size_t)row * k_pitch +
                                          (size_t)kh * D + dt * 32 + lo32);
      }
      kfB[dt][tt] = b.v;
    }

  // block-level q loop bounds (union over the 4 waves' tiles)
  int q_lo = qs, q_hi = qe;
  const int nlast = min(nblk0 + BWD_BN * BWD_WAVES, ke) - 1;
  if (atype == 1 || atype == 3) q_lo = max(q_lo, nblk0 - (ke - qe));
  if (atype == 2 || atype == 3) q_hi = min(q_hi, nlast - (ks - qs) + 1);
  // wave-level bounds
  int wq_lo = qs, wq_hi = qe;
  if (atype == 1 || atype == 3) wq_lo = max(wq_lo, n0 - (ke - qe));
  if (atype == 2 || atype == 3)
    wq_hi = min(wq_hi, (n0 + BWD_BN - 1) - (ks - qs) + 1);

  f32x16 acc_dk[DT], acc_dv[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) {
    acc_dk[dt] = (f32x16)(0.f);
    acc_dv[dt] = (f32x16)(0.f);
  }

  for (int m0 = q_lo; m0 < q_hi; m0 += BWD_BM) {
    // ---- cooperative staging of Q/dO tile (swizzled 16-B chunks) ----
    {
      // 256 threads x 16B; chunks_per_row = D/8; rows_per_pass = 256/(D/8)
      constexpr int CPR = D / 8;           // 16-B chunks per row
      constexpr int RPP = 256 / CPR;       // rows staged per pass
      const int row = tid / CPR;
      const int col = tid % CPR;           // 16-B chunk index
#pragma unroll
      for (int pass = 0; pass < BWD_BM / RPP; ++pass) {
        const int r = pass * RPP + row;
        const int qrow = min(m0 + r, qe - 1);
        const bf16_t* qp = p.q + (size_t)qrow * q_pitch + (size_t)h * D;
        const bf16_t* dp = p.dout + (size_t)qrow * q_pitch + (size_t)h * D;
        const int dst = swz(r, r * ROWB + col * 16);
        *(bf16x8*)((char*)lds_q + dst) = *(const bf16x8*)(qp + col * 8);
        *(bf16x8*)((char*)lds_do + dst) = *(const bf16x8*)(dp + col * 8);
      }
    }
    __syncthreads();

    if (wave_live && m0 + BWD_BM > wq_lo && m0 < wq_hi) {
      const int qrow = m0 + lo32;
      const bool qvalid = qrow < wq_hi && qrow < qe;

      // ---- S^T = K Q^T ; dP^T = V dO^T (q-side B-frags from LDS rows) ----
      f32x16 s = (f32x16)(0.f), dp = (f32x16)(0.f);
#pragma unroll
      for (int dd = 0; dd < DF; ++dd) {
        const int off = swz(lo32, lo32 * ROWB + dd * 32 + hi * 16);
        bf16x8 qf = *(const bf16x8*)((const char*)lds_q + off);
        bf16x8 dof = *(const bf16x8*)((const char*)lds_do + off);
        s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfA[dd], qf, s, 0, 0, 0);
        dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfA[dd], dof, dp, 0, 0, 0);
      }

      const float lse_q = qvalid ? p.lse[(size_t)qrow * p.hq + h] : INFINITY;
      const float dpsum_q = qvalid ? p.dpsum[(size_t)qrow * p.hq + h] : 0.f;
      const bool row_live = qvalid && lse_q != INFINITY && lse_q != -INFINITY;

      float pv[16], dsv[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kk = n0 + crow(r, hi);
        bool ok = row_live && kk < ke;
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
        const float pij = ok ? exp2f(t - lse_q * log2e) : 0.f;
        pv[r] = pij;
        dsv[r] = pij * (dp[r] - dpsum_q) * dscale;
      }

      // ---- dV += P^T dO (P via per-wave LDS transpose; dO cols from LDS) ----
#pragma unroll
      for (int r = 0; r < 16; ++r) lds_p[wave][crow(r, hi)][lo32] = (__bf16)pv[r];
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
        for (int tt = 0; tt < 2; ++tt) {
          Bf b;
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int r = 16 * tt + 8 * hi + e;
            b.u[e] = *(const unsigned short*)(
                (const char*)lds_do + swz(r, r * ROWB + (dt * 32 + lo32) * 2));
          }
          bf16x8 pa = *(const bf16x8*)(&lds_p[wave][lo32][0] + 16 * tt + 8 * hi);
          acc_dv[dt] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, b.v, acc_dv[dt], 0, 0, 0);
        }
      }

      // ---- dK += dS^T Q (dS via LDS transpose; Q cols from LDS) ----
#pragma unroll
      for (int r = 0; r < 16; ++r) lds_p[wave][crow(r, hi)][lo32] = (__bf16)dsv[r];
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
        for (int tt = 0; tt < 2; ++tt) {
          Bf b;
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int r = 16 * tt + 8 * hi + e;
            b.u[e] = *(const unsigned short*)(
                (const char*)lds_q + swz(r, r * ROWB + (dt * 32 + lo32) * 2));
          }
          bf16x8 dsa = *(const bf16x8*)(&lds_p[wave][lo32][0] + 16 * tt + 8 * hi);
          acc_dk[dt] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(dsa, b.v, acc_dk[dt], 0, 0, 0);
        }
      }

      // ---- dQ += dS K (atomicAdd; dS^T -> A-frag in-register) ----
      // one d-tile at a time: keeps the transient accumulator at 16 regs
      // (a 4-tile acc_dq overflowed the unified register file -> scratch)
      {
        bf16x8 dsa0 = cframe_to_afrag(dsv, 0);
        bf16x8 dsa1 = cframe_to_afrag(dsv, 1);
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) {
          f32x16 acc = (f32x16)(0.f);
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dsa0, kfB[dt][0], acc, 0, 0, 0);
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dsa1, kfB[dt][1], acc, 0, 0, 0);
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int qr = m0 + crow(r, hi);
            if (qr >= qe || skip_dq) continue;
            const float val = acc[r];
            if (val != 0.f)
              unsafeAtomicAdd(
                  p.dq + (size_t)qr * q_pitch + (size_t)h * D + dt * 32 + lo32,
                  val);
          }
        }
      }
    }
    __syncthreads();
  }

  // ---- write dK/dV ----
  if (!wave_live || (p.debug_ablate & 2)) return;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kr = n0 + crow(r, hi);
    if (kr >= ke) continue;
    float* dkp = p.dk + (size_t)kr * k_pitch + (size_t)kh * D;
    float* dvp = p.dv + (size_t)kr * k_pitch + (size_t)kh * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      if (acc_dk[dt][r] != 0.f) unsafeAtomicAdd(dkp + dt * 32 + lo32, acc_dk[dt][r]);
      if (acc_dv[dt][r] != 0.f) unsafeAtomicAdd(dvp + dt * 32 + lo32, acc_dv[dt][r]);
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

extern "C" int magi_ffa_bwd(const magi_ffa_bwd_args* a) {
  if (!a || !a->dout || !a->q || !a->k || !a->v || !a->lse) return -1;
  if (a->d != 64 && a->d != 128) return -2;
  if (a->hq % a->hk != 0) return -3;
  if (a->n_ranges <= 0) return 0;

  BwdParams p{};
  p.dout = (const bf16_t*)a->dout;
  p.q = (const bf16_t*)a->q;
  p.k = (const bf16_t*)a->k;
  p.v = (const bf16_t*)a->v;
  p.out = a->out;
  p.lse = a->lse;
  p.dq = a->dq;
  p.dk = a->dk;
  p.dv = a->dv;
  p.dpsum = a->dpsum;
  p.q_ranges = a->q_ranges;
  p.k_ranges = a->k_ranges;
  p.attn_type_map = a->attn_type_map;
  p.hq = a->hq;
  p.hk = a->hk;
  p.gqa = a->hq / a->hk;
  p.scale = a->softmax_scale;
  p.softcap = a->softcap;
  p.total_q = a->total_q;
  p.total_k = a->total_k;
  { const char* e = getenv("MAGI_BWD_ABLATE"); p.debug_ablate = e ? atoi(e) : 0; }

  const int span = BWD_BN * BWD_WAVES;
  const int nblocks = (a->max_seqlen_k + span - 1) / span;
  if (a->n_ranges > 65535) return -5;
  dim3 grid(a->hq, nblocks, (unsigned)a->n_ranges), block(64 * BWD_WAVES);
  hipStream_t s = (hipStream_t)a->stream;
  const bool sc = a->softcap > 0.f;
  if (a->d == 64) {
    if (sc)
      hipLaunchKernelGGL((ffa_bwd_kernel<64, true>), grid, block, 0, s, p);
    else
      hipLaunchKernelGGL((ffa_bwd_kernel<64, false>), grid, block, 0, s, p);
  } else {
    if (sc)
      hipLaunchKernelGGL((ffa_bwd_kernel<128, true>), grid, block, 0, s, p);
    else
      hipLaunchKernelGGL((ffa_bwd_kernel<128, false>), grid, block, 0, s, p);
  }
  return (int)hipGetLastError();
}
