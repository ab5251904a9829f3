// ffa_index.hip — MI355X-native index-attention (token-gather) FORWARD kernel.
//
// Reference surface: flex_flash_attn.py:1358-1390 (index_attn_indices direct
// path) + tests/test_attn/test_index_attn.py semantics:
//   - indices_2d [total_q, max_topk] int32: GLOBAL K row ids attended by q
//     token row i, shared by ALL hq query heads of that row; -1 = padding,
//     contiguous at the tail (the kernel finds the count itself).
//   - KV heads are folded into the K row dimension (hk == 1): global id for
//     (batch b, token t, kv-head h) is (b*S_kv + t)*NHK + h
//     (utils/sparse_utils.py:534-574).
//   - forward only (reference backward receives None for index_attn,
//     flex_flash_attn.py:950-954).
//
// MI355X mapping (NOT the reference's SM90 kBlockM/PackGQA scheme): the MFMA
// M dimension is the token's hq query heads — the reference's PackGQA is the
// *implicit* layout here, since a q row already groups the GQA heads that
// share one index list. K/V tiles are gathered row-by-row straight into LDS
// with lane-indexed global_load_lds (the LDS-DMA path takes an independent
// address per lane, so a gather costs the same issue slots as the contiguous
// stage in ffa_fwd.hip). One workgroup owns one (token, head-block): direct
// epilogue, no locks, no cross-WG reduction.
//
// Compute core (swapped QK^T, softmax layout, defer-max, cvt_pk P rebuild,
// tr16 V fragments) is the same design as ffa_fwd.hip — see the notes there.

#include <hip/hip_runtime.h>
#include <math.h>

#include "../../include/magi_ffa.h"

#define IDX_BN 32  // k rows per inner MFMA tile

using f32x16 = __attribute__((ext_vector_type(16))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using bf16_t = __bf16;

#define DEV_INLINE __device__ __forceinline__

DEV_INLINE float idx_warp_xor32(float v) { return __shfl_xor(v, 32, 64); }

DEV_INLINE unsigned idx_pack_bf16_pair(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

DEV_INLINE float idx_fast_exp2(float x) { return __builtin_amdgcn_exp2f(x); }

DEV_INLINE int idx_crow(int r, int hi) { return (r & 3) + 8 * (r >> 2) + 4 * hi; }

DEV_INLINE bf16x8 idx_tr16_frag(int a0, int a1) {
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

struct IdxParams {
  const bf16_t* q;
  const bf16_t* k;
  const bf16_t* v;
  float* out_f32;
  bf16_t* out_bf16;
  float* lse;
  const int* idx2d;  // [total_q, max_topk]
  int max_topk;
  int hq;
  float scale;
  float softcap;
  long long total_q, total_k;
};

template <int D, bool HAS_SOFTCAP, bool OUT_BF16, int WAVES>
__global__ __launch_bounds__(64 * WAVES, 8 / WAVES) void ffa_index_fwd_kernel(
    IdxParams p) {
  constexpr int DF = D / 16;
  constexpr int DT = D / 32;

  const long long tok = blockIdx.x;
  const int h0 = blockIdx.y * (32 * WAVES);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo32 = lane & 31;
  const int hi = lane >> 5;

  // ---- valid-count: the -1 padding is contiguous at the tail, so a
  // wave-uniform binary search (same address every lane -> broadcast loads)
  // finds it in log2(max_topk) dwords ----
  const int* irow = p.idx2d + tok * (long long)p.max_topk;
  int count = p.max_topk;
  if (irow[p.max_topk - 1] < 0) {
    int lo = 0, hic = p.max_topk - 1;
    while (lo < hic) {
      const int mid = (lo + hic) >> 1;
      if (irow[mid] < 0)
        hic = mid;
      else
        lo = mid + 1;
    }
    count = lo;
  }
  if (count == 0) return;  // out zero-init + lse -inf preset by the host

  const int q0 = h0 + wave * 32;       // first query head of this wave
  const int qh = q0 + lo32;            // this lane's query head
  const bool qvalid = qh < p.hq;
  const bool qvalid_any = q0 < p.hq;
  const int qclamp = qvalid ? qh : (p.hq - 1);

  const float sl2 = HAS_SOFTCAP ? p.softcap * 1.4426950408889634f
                                : p.scale * 1.4426950408889634f;
  const float cap_pre = HAS_SOFTCAP ? p.scale / p.softcap : 0.f;

  constexpr int ROWB = D * 2;          // bf16 row bytes (D=64 -> 128, D=128 -> 256)
  constexpr int ROWE = ROWB / 2;
  constexpr int SW32M = ROWB / 32 - 1;
  auto swz = [](int row, int byte_off) {
    return byte_off ^ ((row & SW32M) << 5);
  };

  // Staged-image depth scales with the wave count: small-ratio workgroups
  // (1-2 waves, hq<=64) keep 32-row buffers so 4-5 WGs co-reside per CU
  // (the 64-row image at 64 KB capped LDS occupancy at 2 WGs -> 0.5-1
  // waves/SIMD, measured 88 TF at hq=32); 4-wave WGs keep the 64-row image
  // (2 WGs/CU, 2 waves/SIMD, one barrier per 64 gathered rows).
  constexpr int IDX_KITER = (WAVES >= 4) ? 64 : 32;
  __shared__ __attribute__((aligned(16))) char ismem[2 * 2 * IDX_KITER * ROWB];
  auto lds_k = [&](int buf) -> __bf16* {
    return (__bf16*)(ismem + (2 * buf) * IDX_KITER * ROWB);
  };
  auto lds_v = [&](int buf) -> __bf16* {
    return (__bf16*)(ismem + (2 * buf + 1) * IDX_KITER * ROWB);
  };

  // Q fragments in registers
  bf16x8 qf[DF];
  {
    const bf16_t* qp = p.q + (tok * p.hq + qclamp) * (long long)D;
#pragma unroll
    for (int dd = 0; dd < DF; ++dd)
      qf[dd] = *(const bf16x8*)(qp + dd * 16 + hi * 8);
  }

  float m_run = -INFINITY;
  float l_run = 0.f;
  f32x16 acc_o[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) acc_o[dt] = (f32x16)(0.f);

  constexpr int ROWS_PER_GLDS = 1024 / ROWB;
  static_assert(IDX_KITER / WAVES >= ROWS_PER_GLDS, "stage rows per wave");
  constexpr int GLDS_PER_WAVE = (IDX_KITER / WAVES) / ROWS_PER_GLDS;
  // Gathered stage: global row = irow[position], clamped for safety. The
  // staged positions never exceed max_topk-1 (64 | max_topk and the loop
  // rounds count up to 64).
  auto stage_glds = [&](int buf, int n0x) {
#pragma unroll
    for (int gi = 0; gi < GLDS_PER_WAVE; ++gi) {
      const int r0 = (IDX_KITER / WAVES) * wave + ROWS_PER_GLDS * gi;
      const int r = r0 + lane / (ROWB / 16);
      const int c = lane % (ROWB / 16);
      long long kr = irow[n0x + r];
      if (kr < 0 || kr >= p.total_k) kr = 0;  // pad/garbage: any in-bounds row
      const int cs = c ^ ((r & SW32M) << 1);
      const int csw = cs * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.k + kr * (long long)D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_k(buf)[r0 * ROWE],
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              p.v + kr * (long long)D + csw),
          (__attribute__((address_space(3))) unsigned int*)&lds_v(buf)[r0 * ROWE],
          16, 0, 0);
    }
  };

  auto sub_body = [&](int ns, const __bf16* lkb, const __bf16* lvb) {
    if (!(ns < count && qvalid_any)) return;

    f32x16 s = (f32x16)(0.f);
#pragma unroll
    for (int dd = 0; dd < DF; ++dd) {
      bf16x8 kf = *(const bf16x8*)(
          (const char*)lkb + swz(lo32, lo32 * ROWB + dd * 32 + hi * 16));
      s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[dd], s, 0, 0, 0);
    }

    // ---- mask (position >= count) + scale into exp2 domain ----
    const bool interior = (q0 + 31 < p.hq) && (ns + IDX_BN <= count);
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
        const int kk = ns + idx_crow(r, hi);
        const bool ok = qvalid && (kk < count);
        float sv = s[r];
        if (HAS_SOFTCAP) sv = tanhf(sv * cap_pre);
        t[r] = ok ? sv * sl2 : -INFINITY;
        mx = fmaxf(mx, t[r]);
      }
    }
    mx = fmaxf(mx, idx_warp_xor32(mx));

    // ---- defer-max (see ffa_fwd.hip) ----
    const float m_new = fmaxf(m_run, mx);
    const bool need_rescale =
        (m_new > m_run + 8.f) || (m_run == -INFINITY && m_new != -INFINITY);
    float m_use;
    float alpha;
    if (!__any(need_rescale)) {
      m_use = (m_run == -INFINITY) ? 0.f : m_run;
      alpha = 1.f;
    } else {
      m_use = (m_new == -INFINITY) ? 0.f : m_new;
      alpha = (m_run == -INFINITY) ? 0.f : idx_fast_exp2(m_run - m_use);
      m_run = m_new;
    }

    float pr[16];
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      pr[r] = idx_fast_exp2(t[r] - m_use);
      psum += pr[r];
    }
    l_run = l_run * alpha + (psum + idx_warp_xor32(psum));

    if (__any(alpha != 1.f)) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int src = idx_crow(r, hi);
        const float aq = __uint_as_float(
            __builtin_amdgcn_ds_bpermute(src << 2, __float_as_uint(alpha)));
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) acc_o[dt][r] *= aq;
      }
    }

    // ---- P -> bf16 A-fragments ----
    bf16x8 pa[2];
#pragma unroll
    for (int tt = 0; tt < 2; ++tt) {
      unsigned c0 = idx_pack_bf16_pair(pr[8 * tt + 0], pr[8 * tt + 1]);
      unsigned c1 = idx_pack_bf16_pair(pr[8 * tt + 2], pr[8 * tt + 3]);
      unsigned c2 = idx_pack_bf16_pair(pr[8 * tt + 4], pr[8 * tt + 5]);
      unsigned c3 = idx_pack_bf16_pair(pr[8 * tt + 6], pr[8 * tt + 7]);
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

    // ---- PV ----
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
          bf16x8 bv = idx_tr16_frag(rb0 + (dcol ^ sw0), rb1 + (dcol ^ sw1));
          acc_o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[tt], bv,
                                                              acc_o[dt], 0, 0, 0);
        }
      }
    }
  };

  int cur = 0;
  stage_glds(0, 0);
  for (int n0 = 0; n0 < count; n0 += IDX_KITER) {
    __syncthreads();  // buf[cur] glds drained
    if (n0 + IDX_KITER < count) stage_glds(cur ^ 1, n0 + IDX_KITER);
#pragma unroll
    for (int sub = 0; sub < IDX_KITER / IDX_BN; ++sub) {
      const int ns = n0 + sub * IDX_BN;
      if (ns >= count) break;  // count is block-uniform
      sub_body(ns, lds_k(cur) + sub * IDX_BN * ROWE,
               lds_v(cur) + sub * IDX_BN * ROWE);
    }
    cur ^= 1;
  }

  // ======================= epilogue (direct store) =======================
  const float lse_new =
      (l_run > 0.f) ? (m_run + __log2f(l_run)) * 0.6931471805599453f : -INFINITY;
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;

  if (l_run > 0.f && qvalid && hi == 0)
    p.lse[tok * p.hq + qh] = lse_new;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int src = idx_crow(r, hi);
    const float ilq = __uint_as_float(
        __builtin_amdgcn_ds_bpermute(src << 2, __float_as_uint(inv_l)));
    const int qr = q0 + src;
    if (qr >= p.hq || ilq <= 0.f) continue;
    const long long base = (tok * p.hq + qr) * (long long)D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      const float val = acc_o[dt][r] * ilq;
      if (OUT_BF16)
        p.out_bf16[base + dt * 32 + lo32] = (bf16_t)val;
      else
        p.out_f32[base + dt * 32 + lo32] = val;
    }
  }
}

// ------------------------------------------------------------------
// launcher
// ------------------------------------------------------------------
template <int D, int W>
static int launch_index_d(const magi_ffa_index_args* a, const IdxParams& p,
                          dim3 grid, hipStream_t stream) {
  const bool sc = a->softcap > 0.f;
  const bool obf16 = !a->out_is_fp32;
  dim3 block(64 * W);
#define ILAUNCH(SC, OB)                                                      \
  hipLaunchKernelGGL((ffa_index_fwd_kernel<D, SC, OB, W>), grid, block, 0, \
                     stream, p)
  if (sc) {
    if (obf16) ILAUNCH(true, true);
    else ILAUNCH(true, false);
  } else {
    if (obf16) ILAUNCH(false, true);
    else ILAUNCH(false, false);
  }
#undef ILAUNCH
  return (int)hipGetLastError();
}

extern "C" int magi_ffa_fwd_index(const magi_ffa_index_args* a) {
  if (!a || !a->q || !a->k || !a->v || !a->out || !a->lse || !a->indices_2d)
    return -1;
  if (a->d != 64 && a->d != 128) return -2;
  if (a->hk != 1) return -3;  // kv heads are folded into K rows (see header)
  if (a->max_topk <= 0 || (a->max_topk & 63) != 0) return -4;
  if (a->total_q <= 0) return 0;

  IdxParams p;
  p.q = (const bf16_t*)a->q;
  p.k = (const bf16_t*)a->k;
  p.v = (const bf16_t*)a->v;
  p.out_f32 = (float*)a->out;
  p.out_bf16 = (bf16_t*)a->out;
  p.lse = a->lse;
  p.idx2d = a->indices_2d;
  p.max_topk = a->max_topk;
  p.hq = a->hq;
  p.scale = a->softmax_scale;
  p.softcap = a->softcap;
  p.total_q = a->total_q;
  p.total_k = a->total_k;

  // head-block = 32*W query heads; pick W so one wave-set covers hq (DiT
  // ratio 128 -> 4 waves; small-ratio MHA shapes keep lanes busy at W=1)
  int w = (a->hq > 64) ? 4 : (a->hq > 32 ? 2 : 1);
  const int span = 32 * w;
  const unsigned hblocks = (unsigned)((a->hq + span - 1) / span);
  dim3 grid((unsigned)a->total_q, hblocks, 1);
  hipStream_t stream = (hipStream_t)a->stream;
  if (a->d == 64) {
    if (w == 4) return launch_index_d<64, 4>(a, p, grid, stream);
    if (w == 2) return launch_index_d<64, 2>(a, p, grid, stream);
    return launch_index_d<64, 1>(a, p, grid, stream);
  }
  if (w == 4) return launch_index_d<128, 4>(a, p, grid, stream);
  if (w == 2) return launch_index_d<128, 2>(a, p, grid, stream);
  return launch_index_d<128, 1>(a, p, grid, stream);
}
