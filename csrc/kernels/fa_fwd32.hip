// FlashAttention-2 forward, 32x32-MFMA variant (gfx950).
//
// Reference role: paddle/phi/kernels/gpu/flash_attn_kernel.cu:41 (thin
// dispatch into dynloaded libflashattn) -- here the FA2 algorithm is
// implemented natively for CDNA4 instead of vendoring a library.
//
// Design (guide §B "fused attention prefill" 8-warp ladder, re-derived for
// our wave-per-32-q-rows layout):
//   * mfma_f32_32x32x16_bf16; each wave owns 32 q rows; 4 waves/block
//     -> 128-row q blocks; KV tiles of 64.
//   * SWAPPED QK^T: S^T = mfma(K, Q) puts a full q-COLUMN of scores in
//     each lane (C layout col = lane&31 = q), so the online softmax is
//     per-lane scalar + ONE shfl_xor(32) -- no 16-lane reductions.
//   * P stays in registers: the PV A-fragments are assembled with packed
//     bf16 pairs + 2x permlane32_swap per k-step (guide T12) -- no P LDS
//     round-trip, no lgkmcnt(0) stall.
//   * K staged via async global_load_lds with pre-swizzled source (m173);
//     V^T staged with bank-rotated scalar writes.
//   * defer-max rescale (T13), boundary-only masking.
// C/D layout for 32x32 (measured, guide §3): col=lane&31,
// row=(reg&3)+8*(reg>>2)+4*(lane>>5).
#include "common.h"
#include "api.h"

namespace pa {

typedef __attribute__((ext_vector_type(16))) float floatx16;
typedef __attribute__((ext_vector_type(2))) int intx2;

__device__ __forceinline__ floatx16 mfma32_bf16(shortx8 a, shortx8 b, floatx16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

__device__ __forceinline__ unsigned lds_off32(unsigned row, unsigned col_bytes,
                                              unsigned row_stride) {
  return row * row_stride + (col_bytes ^ ((row & 7u) << 4));
}

// 16-granule variant for 256 B rows (D=128 tiles): fragment reads pull 32
// DIFFERENT rows at one column, so an 8-way spread leaves 4-way bank
// conflicts (measured ~1 conflict-cycle/busy-cycle, gpurun fa_pmc.csv);
// 16 granules reach the b128 2-way floor.
__device__ __forceinline__ unsigned lds_off32w(unsigned row, unsigned col_bytes,
                                               unsigned row_stride, unsigned mask) {
  return row * row_stride + (col_bytes ^ ((row & mask) << 4));
}

__device__ __forceinline__ int pack_bf2(float a, float b) {
  unsigned lo = (unsigned short)f2bf(a);
  unsigned hi = (unsigned short)f2bf(b);
  return (int)(lo | (hi << 16));
}

// VARLEN: packed ragged batch (flash_attn_unpadded).  Sq/Skv carry the
// TOTAL q/k token counts; cu_q/cu_k are [nseq+1] prefix sums and bmap
// holds (seq, q0-within-seq) per grid.x block.  Causal is bottom-right
// aligned (kv <= q + Skv_seq - Sq_seq).  lse layout [H, total_q].
template <int D, bool CAUSAL, bool MASKED, bool DROPOUT, bool VARLEN = false>
__launch_bounds__(256, 2)
__global__ void fa_fwd32_kernel(const short* __restrict__ qg, const short* __restrict__ kg,
                                const short* __restrict__ vg, short* __restrict__ og,
                                float* __restrict__ lseg, int B, int H, int HKV,
                                int Sq, int Skv, float scale,
                                long long q_sb, long long q_sh, long long q_ss,
                                long long k_sb, long long k_sh, long long k_ss,
                                long long o_sb, long long o_sh, long long o_ss,
                                const short* __restrict__ maskg, long long m_sb,
                                long long m_sh, long long m_sq,
                                float pdrop, unsigned long long rseed,
                                unsigned long long roffset,
                                const int* __restrict__ cu_q = nullptr,
                                const int* __restrict__ cu_k = nullptr,
                                const int* __restrict__ bmap = nullptr) {
  constexpr int NW = 4;            // waves per block, 32 q rows each
  constexpr int NT = NW * 64;
  constexpr int QB = NW * 32;      // 128 q rows per block
  constexpr int KB = 64;           // kv tile
  constexpr int NKS = D / 16;      // q/k k-steps over head dim (mfma K=16)
  constexpr int NDT = D / 32;      // output d tiles (32 wide)
  constexpr unsigned K_RS = D * 2;
  constexpr unsigned VT_RS = KB * 2;
  // double-buffered: stage tile t+1 while the MFMAs chew tile t
  __shared__ char k_lds[2][KB * D * 2];
  __shared__ char vt_lds[2][D * KB * 2];

  // block index on the SLOW grid dim (y) so causal longest-first ordering
  // is global: the dispatcher walks x fastest, so a length-ordered y makes
  // the drain tail all-short blocks instead of one of each length.
  const int qblk = blockIdx.y;
  const int bh = blockIdx.x;
  int b = bh / H, h = bh % H;
  int q0, Sq_e = Sq, Skv_e = Skv, cdelta = 0, qglob0 = 0;
  long long qbase, kbase, obase;
  if (VARLEN) {
    b = 0; h = bh;
    const int seq = bmap[2 * qblk];
    q0 = bmap[2 * qblk + 1];
    const int qs0 = cu_q[seq], ks0 = cu_k[seq];
    Sq_e = cu_q[seq + 1] - qs0;
    Skv_e = cu_k[seq + 1] - ks0;
    cdelta = Skv_e - Sq_e;
    qglob0 = qs0;
    const int hkv_ = h / (H / HKV);
    qbase = (long long)qs0 * q_ss + (long long)h * q_sh;
    kbase = (long long)ks0 * k_ss + (long long)hkv_ * k_sh;
    obase = (long long)qs0 * o_ss + (long long)h * o_sh;
  } else {
    const int hkv_ = h / (H / HKV);
    q0 = (CAUSAL ? ((int)gridDim.y - 1 - qblk) : qblk) * QB;
    qbase = (long long)b * q_sb + (long long)h * q_sh;
    kbase = (long long)b * k_sb + (long long)hkv_ * k_sh;
    obase = (long long)b * o_sb + (long long)h * o_sh;
  }

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wq = tid >> 6;
  const int l32 = lane & 31;       // q column (C layout)
  const int hi = lane >> 5;        // half-wave id
  const int q0w = q0 + wq * 32;

  // Q fragments (B operand of the swapped QK^T): lane holds
  // Q[q0w + l32][16*ks + 8*hi .. +7].  Pre-scaled by softmax scale ONCE
  // here -- saves one VALU mul per score element in the kv loop (the
  // softmax block is what keeps this kernel VALU-issue-bound).
  shortx8 qf[NKS];
  {
    int row = q0w + l32;
    bool ok = row < Sq_e;
#pragma unroll
    for (int ks = 0; ks < NKS; ++ks) {
      if (ok) {
        shortx8 raw = *reinterpret_cast<const shortx8*>(
            qg + qbase + (long long)row * q_ss + ks * 16 + hi * 8);
#pragma unroll
        for (int i = 0; i < 8; ++i) qf[ks][i] = f2bf(bf2f(raw[i]) * scale);
      } else {
        for (int i = 0; i < 8; ++i) qf[ks][i] = 0;
      }
    }
  }

  floatx16 oacc[NDT];
#pragma unroll
  for (int dt = 0; dt < NDT; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) oacc[dt][r] = 0.f;
  float m_run = -INFINITY, l_run = 0.f;   // stats for q column l32

  const int kv_end = CAUSAL ? min(Skv_e, q0 + QB + cdelta) : Skv_e;

  // K-staging source offsets (pre-swizzled so async DMA lands swizzled rows)
  constexpr unsigned KSWZ = (D >= 128) ? 15u : 7u;
  int k_row[KB * D / (NT * 8)], k_colp[KB * D / (NT * 8)];
#pragma unroll
  for (int it = 0; it < KB * D / (NT * 8); ++it) {
    int flat = it * NT * 8 + tid * 8;
    int row = flat / D, col = flat % D;
    k_row[it] = row;
    k_colp[it] = col ^ ((row & (int)KSWZ) << 3);
  }

  auto stage = [&](int buf, int kv0) {
    // ---- stage K [KB][D] (async, swizzled) + V^T [D][KB] ------------------
    if (kv0 + KB <= Skv_e) {
#pragma unroll
      for (int it = 0; it < KB * D / (NT * 8); ++it) {
        const short* src = kg + kbase + (long long)(kv0 + k_row[it]) * k_ss + k_colp[it];
        char* dst = k_lds[buf] + it * NT * 16 + (tid >> 6) * 64 * 16;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)src,
            (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
      }
    } else {
      for (int flat = tid * 8; flat < KB * D; flat += NT * 8) {
        int row = flat / D, col = flat % D;
        shortx8 val;
        if (kv0 + row < Skv_e)
          val = *reinterpret_cast<const shortx8*>(kg + kbase + (long long)(kv0 + row) * k_ss + col);
        else
          for (int i = 0; i < 8; ++i) val[i] = 0;
        *reinterpret_cast<shortx8*>(k_lds[buf] + lds_off32w(row, col * 2, K_RS, KSWZ)) = val;
      }
    }
    {
      const int rot = tid & 7;
      for (int flat = tid * 8; flat < KB * D; flat += NT * 8) {
        int row = flat / D, col = flat % D;  // row=kv, col=d
        shortx8 val;
        if (kv0 + row < Skv_e)
          val = *reinterpret_cast<const shortx8*>(vg + kbase + (long long)(kv0 + row) * k_ss + col);
        else
          for (int i = 0; i < 8; ++i) val[i] = 0;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          *reinterpret_cast<short*>(vt_lds[buf] + lds_off32(col + i, row * 2, VT_RS)) = val[i];
        }
      }
    }
  };

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int kv0 = 0; kv0 < kv_end; kv0 += KB) {
    if (kv0 + KB < kv_end) stage(cur ^ 1, kv0 + KB);

    // ---- S^T = K Q^T : 2 kv tiles of 32 -----------------------------------
    floatx16 st[2];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
#pragma unroll
      for (int r = 0; r < 16; ++r) st[t][r] = 0.f;
#pragma unroll
      for (int ks = 0; ks < NKS; ++ks) {
        shortx8 kf = *reinterpret_cast<const shortx8*>(
            k_lds[cur] + lds_off32w(t * 32 + l32, (ks * 16 + hi * 8) * 2, K_RS, KSWZ));
        st[t] = mfma32_bf16(kf, qf[ks], st[t]);
      }
    }

    // ---- mask + scale + per-lane online softmax ---------------------------
    const int q_abs = q0w + l32;
    if (MASKED && q_abs < Sq_e) {
      // additive mask [B, 1|H, Sq, Skv]; reg groups are 4 consecutive kv
      const short* mrow = maskg + (long long)b * m_sb + (long long)h * m_sh
                        + (long long)q_abs * m_sq;
#pragma unroll
      for (int t = 0; t < 2; ++t)
#pragma unroll
        for (int g = 0; g < 4; ++g) {
          long long kvb = kv0 + 32 * t + 8 * g + 4 * hi;
          if (kvb + 3 < Skv_e) {
            shortx4 mv = *reinterpret_cast<const shortx4*>(mrow + kvb);
#pragma unroll
            for (int j = 0; j < 4; ++j) st[t][4 * g + j] += bf2f(mv[j]);
          } else {
#pragma unroll
            for (int j = 0; j < 4; ++j)
              if (kvb + j < Skv_e) st[t][4 * g + j] += bf2f(mrow[kvb + j]);
          }
        }
    }
    float mx = -INFINITY;
    const bool boundary = (kv0 + KB > Skv_e) || (CAUSAL && kv0 + KB > q_abs + cdelta);
    if (boundary) {
#pragma unroll
      for (int t = 0; t < 2; ++t)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv_abs = kv0 + 32 * t + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float v = st[t][r];                 // Q pre-scaled at load
          if (kv_abs >= Skv_e || (CAUSAL && kv_abs > q_abs + cdelta)) v = -INFINITY;
          st[t][r] = v;
          mx = fmaxf(mx, v);
        }
    } else {
#pragma unroll
      for (int t = 0; t < 2; ++t)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          mx = fmaxf(mx, st[t][r]);           // Q pre-scaled at load
        }
    }
    mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
    const float THR = 8.f;
    // per-lane stats are divergent; compute corr for every lane so the
    // broadcast shfl below is wave-uniform-safe, and vote before paying
    // for the O rescale
    const bool grow = (mx > m_run + THR) || (m_run == -INFINITY);
    float m_new = grow ? fmaxf(m_run, mx) : m_run;
    float corr = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
    if (__any(grow && m_run != -INFINITY)) {
      l_run *= corr;
#pragma unroll
      for (int dt = 0; dt < NDT; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int rq = (r & 3) + 8 * (r >> 2) + 4 * hi;
          float c = __shfl(corr, rq, 64);
          oacc[dt][r] *= c;
        }
    }
    m_run = m_new;
    float psum = 0.f;
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float p = (st[t][r] == -INFINITY) ? 0.f : __expf(st[t][r] - m_run);
        st[t][r] = p;
        psum += p;
      }
    psum += __shfl_xor(psum, 32, 64);
    l_run += psum;
    if (DROPOUT) {
      // Philox keep-mask on P (normalizer above uses undropped P)
      const float inv_keep = 1.f / (1.f - pdrop);
      const unsigned thr24 = (unsigned)(pdrop * 16777216.f);
      const long long bh_ll = (long long)b * H + h;
#pragma unroll
      for (int t = 0; t < 2; ++t)
#pragma unroll
        for (int g = 0; g < 4; ++g) {
          long long kvb = kv0 + 32 * t + 8 * g + 4 * hi;
          long long elem = (bh_ll * Sq + (q_abs + qglob0)) * (long long)Skv + kvb;
          uintx4 rv = philox10(rseed, roffset + (unsigned long long)(elem >> 2));
#pragma unroll
          for (int j = 0; j < 4; ++j)
            st[t][4 * g + j] = ((rv[j] >> 8) >= thr24)
                                   ? st[t][4 * g + j] * inv_keep : 0.f;
        }
    }

    // ---- O += P V : assemble P A-frags in-register ------------------------
#pragma unroll
    for (int ks = 0; ks < KB / 16; ++ks) {
      const int t = ks >> 1;
      const int kp = ks & 1;
      // STATIC register indices only (rule #20: a runtime `hi` index into
      // the v16 accumulator becomes a 16-way cndmask tree); pack all four
      // candidate words, then one select each by hi.
      const int b0 = 8 * kp;
      int P0 = pack_bf2(st[t][b0 + 0], st[t][b0 + 1]);
      int P1 = pack_bf2(st[t][b0 + 2], st[t][b0 + 3]);
      int P2 = pack_bf2(st[t][b0 + 4], st[t][b0 + 5]);
      int P3 = pack_bf2(st[t][b0 + 6], st[t][b0 + 7]);
      int O1 = hi ? P2 : P0;             // own reg-group words
      int O2 = hi ? P3 : P1;
      int S1 = hi ? P0 : P2;             // supply (partner's) words
      int S2 = hi ? P1 : P3;
      intx2 ra = __builtin_amdgcn_permlane32_swap(S1, S2, false, false);
      intx2 rb = __builtin_amdgcn_permlane32_swap(S2, S1, false, false);
      // lanes<32: partner supply = (ra[1], rb[1]); lanes>=32: (rb[0], ra[0])
      int X1 = hi ? rb[0] : ra[1];
      int X2 = hi ? ra[0] : rb[1];
      // A-frag words in kv order: hi==0 -> {own, partner}; hi==1 -> {partner, own}
      intx4 paw;
      paw[0] = hi ? X1 : O1;
      paw[1] = hi ? X2 : O2;
      paw[2] = hi ? O1 : X1;
      paw[3] = hi ? O2 : X2;
      shortx8 pa = *reinterpret_cast<shortx8*>(&paw);
#pragma unroll
      for (int dt = 0; dt < NDT; ++dt) {
        shortx8 vf = *reinterpret_cast<const shortx8*>(
            vt_lds[cur] + lds_off32(dt * 32 + l32, (ks * 16 + hi * 8) * 2, VT_RS));
        oacc[dt] = mfma32_bf16(pa, vf, oacc[dt]);
      }
    }
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: O /= l, store O + LSE ------------------------------------
  float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int rq = (r & 3) + 8 * (r >> 2) + 4 * hi;
    float il = __shfl(inv_l, rq, 64);
    int row = q0w + rq;
    if (row >= Sq_e) continue;
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt)
      og[obase + (long long)row * o_ss + dt * 32 + l32] = f2bf(oacc[dt][r] * il);
  }
  if (hi == 0 && q0w + l32 < Sq_e)
    lseg[(long long)(b * H + h) * Sq + qglob0 + q0w + l32] =
        (l_run > 0.f) ? m_run + __logf(l_run) : -INFINITY;
}

void flash_attn_fwd32(const void* q, const void* k, const void* v, void* o,
                      float* lse, int64_t b, int64_t h, int64_t hkv, int64_t sq,
                      int64_t skv, int64_t dh, float scale, bool causal,
                      const int64_t* qs, const int64_t* ks, const int64_t* os,
                      const void* mask, const int64_t* ms, float pdrop,
                      uint64_t seed, uint64_t offset, hipStream_t s) {
  dim3 grid((unsigned)(b * h), (unsigned)cdiv((int)sq, 128));
  dim3 blk(256);
  const bool masked = mask != nullptr;
  const bool dropped = pdrop > 0.f;
  static const int64_t zs[3] = {0, 0, 0};
  if (!ms) ms = zs;
#define FAF1(D, C, M, P)                                                        \
  hipLaunchKernelGGL((fa_fwd32_kernel<D, C, M, P>), grid, blk, 0, s,            \
                     (const short*)q, (const short*)k, (const short*)v,         \
                     (short*)o, lse, (int)b, (int)h, (int)hkv, (int)sq,         \
                     (int)skv, scale, qs[0], qs[1], qs[2],                      \
                     ks[0], ks[1], ks[2], os[0], os[1], os[2],                  \
                     (const short*)mask, ms[0], ms[1], ms[2], pdrop, seed, offset)
#define FAF(D, C)                                                               \
  do {                                                                          \
    if (masked && dropped) FAF1(D, C, true, true);                              \
    else if (masked)       FAF1(D, C, true, false);                             \
    else if (dropped)      FAF1(D, C, false, true);                             \
    else                   FAF1(D, C, false, false);                            \
  } while (0)
  if (dh == 128) { if (causal) FAF(128, true); else FAF(128, false); }
  else           { if (causal) FAF(64, true);  else FAF(64, false); }
#undef FAF
#undef FAF1
}



// ---------------------------------------------------------------------------
// backward dQ, 32x32 variant: same swapped layout as the forward -- S^T and
// dP^T put the q-COLUMN in each lane (per-lane lse/delta, no broadcasts),
// and the dS A-fragments for dQ += dS.K are assembled with the identical
// static-pack + permlane32_swap exchange (no LDS round trip).
// K/V staged natural via async DMA; K^T staged with rotated writes for the
// dQ B-fragments.
// ---------------------------------------------------------------------------
namespace {
constexpr int DQ_NW = 4;
}

template <int D, bool CAUSAL, bool MASKED, bool DROPOUT, bool VARLEN = false>
__launch_bounds__(256, 2)
__global__ void fa_bwd_dq32_kernel(const short* __restrict__ dog, const short* __restrict__ qg,
                                   const short* __restrict__ kg, const short* __restrict__ vg,
                                   const float* __restrict__ lseg, const float* __restrict__ deltag,
                                   short* __restrict__ dqg, int B, int H, int Sq, int Skv,
                                   float scale,
                                   long long q_sb, long long q_sh, long long q_ss,
                                   long long k_sb, long long k_sh, long long k_ss,
                                   long long do_sb, long long do_sh, long long do_ss,
                                   long long dq_sb, long long dq_sh, long long dq_ss,
                                   const short* __restrict__ maskg, long long m_sb,
                                   long long m_sh, long long m_sq,
                                   float pdrop, unsigned long long rseed,
                                   unsigned long long roffset,
                                   const int* __restrict__ cu_q = nullptr,
                                   const int* __restrict__ cu_k = nullptr,
                                   const int* __restrict__ bmap = nullptr) {
  constexpr int NW = DQ_NW;
  constexpr int NT = NW * 64;
  constexpr int QB = NW * 32;       // 128 q rows / block
  constexpr int KB = 64;
  constexpr int NKS = D / 16;
  constexpr int NDT = D / 32;
  constexpr unsigned NAT_RS = D * 2;
  constexpr unsigned KT_RS = KB * 2;
  __shared__ char k_lds[KB * D * 2];
  __shared__ char v_lds[KB * D * 2];
  __shared__ char kt_lds[D * KB * 2];

  const int qblk = blockIdx.y;      // slow dim: global longest-first causal order
  const int bh = blockIdx.x;
  int b = bh / H, h = bh % H;
  int q0, Sq_e = Sq, Skv_e = Skv, cdelta = 0, qglob0 = 0;
  long long qbase, dobase, dqbase, kvbase;
  if (VARLEN) {
    b = 0; h = bh;
    const int seq = bmap[2 * qblk];
    q0 = bmap[2 * qblk + 1];
    const int qs0 = cu_q[seq], ks0 = cu_k[seq];
    Sq_e = cu_q[seq + 1] - qs0;
    Skv_e = cu_k[seq + 1] - ks0;
    cdelta = Skv_e - Sq_e;
    qglob0 = qs0;
    qbase = (long long)qs0 * q_ss + (long long)h * q_sh;
    dobase = (long long)qs0 * do_ss + (long long)h * do_sh;
    dqbase = (long long)qs0 * dq_ss + (long long)h * dq_sh;
    kvbase = (long long)ks0 * k_ss + (long long)h * k_sh;
  } else {
    q0 = (CAUSAL ? ((int)gridDim.y - 1 - qblk) : qblk) * QB;   // longest-first
    qbase = (long long)b * q_sb + (long long)h * q_sh;
    dobase = (long long)b * do_sb + (long long)h * do_sh;
    dqbase = (long long)b * dq_sb + (long long)h * dq_sh;
    kvbase = (long long)b * k_sb + (long long)h * k_sh;
  }
  const long long lse_base = ((long long)bh) * Sq + qglob0;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wq = tid >> 6;
  const int l32 = lane & 31;
  const int hi = lane >> 5;
  const int q0w = q0 + wq * 32;

  // Q and dO as B-fragments (lane holds the q column l32); Q pre-scaled
  // by the softmax scale so the exp loop skips a per-element mul
  shortx8 qf[NKS], dof[NKS];
  {
    int row = q0w + l32;
    bool ok = row < Sq_e;
#pragma unroll
    for (int ks = 0; ks < NKS; ++ks) {
      if (ok) {
        shortx8 raw = *reinterpret_cast<const shortx8*>(
            qg + qbase + (long long)row * q_ss + ks * 16 + hi * 8);
#pragma unroll
        for (int i = 0; i < 8; ++i) qf[ks][i] = f2bf(bf2f(raw[i]) * scale);
        dof[ks] = *reinterpret_cast<const shortx8*>(
            dog + dobase + (long long)row * do_ss + ks * 16 + hi * 8);
      } else {
        for (int i = 0; i < 8; ++i) { qf[ks][i] = 0; dof[ks][i] = 0; }
      }
    }
  }
  const int q_abs = q0w + l32;
  const float lse_v = (q_abs < Sq_e) ? lseg[lse_base + q_abs] : 1e30f;
  const float delta_v = (q_abs < Sq_e) ? deltag[lse_base + q_abs] : 0.f;

  floatx16 dq_acc[NDT];
#pragma unroll
  for (int dt = 0; dt < NDT; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[dt][r] = 0.f;

  // DMA staging maps
  constexpr unsigned KSWZ = (D >= 128) ? 15u : 7u;
  int n_row[KB * D / (NT * 8)], n_colp[KB * D / (NT * 8)];
#pragma unroll
  for (int it = 0; it < KB * D / (NT * 8); ++it) {
    int flat = it * NT * 8 + tid * 8;
    int row = flat / D, col = flat % D;
    n_row[it] = row;
    n_colp[it] = col ^ ((row & (int)KSWZ) << 3);
  }

  const int kv_end = CAUSAL ? min(Skv_e, q0 + QB + cdelta) : Skv_e;
  for (int kv0 = 0; kv0 < kv_end; kv0 += KB) {
    if (kv0 + KB <= Skv_e) {
#pragma unroll
      for (int it = 0; it < KB * D / (NT * 8); ++it) {
        const short* ksrc = kg + kvbase + (long long)(kv0 + n_row[it]) * k_ss + n_colp[it];
        const short* vsrc = vg + kvbase + (long long)(kv0 + n_row[it]) * k_ss + n_colp[it];
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)ksrc,
            (__attribute__((address_space(3))) unsigned int*)(k_lds + it * NT * 16 + (tid >> 6) * 64 * 16),
            16, 0, 0);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)vsrc,
            (__attribute__((address_space(3))) unsigned int*)(v_lds + it * NT * 16 + (tid >> 6) * 64 * 16),
            16, 0, 0);
      }
      const int rot = tid & 7;
      for (int flat = tid * 8; flat < KB * D; flat += NT * 8) {
        int row = flat / D, col = flat % D;
        shortx8 kv_ = *reinterpret_cast<const shortx8*>(
            kg + kvbase + (long long)(kv0 + row) * k_ss + col);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          *reinterpret_cast<short*>(kt_lds + lds_off32(col + i, row * 2, KT_RS)) = kv_[i];
        }
      }
    } else {
      for (int flat = tid * 8; flat < KB * D; flat += NT * 8) {
        int row = flat / D, col = flat % D;
        shortx8 kv_, vv;
        if (kv0 + row < Skv_e) {
          kv_ = *reinterpret_cast<const shortx8*>(kg + kvbase + (long long)(kv0 + row) * k_ss + col);
          vv = *reinterpret_cast<const shortx8*>(vg + kvbase + (long long)(kv0 + row) * k_ss + col);
        } else {
          for (int i = 0; i < 8; ++i) { kv_[i] = 0; vv[i] = 0; }
        }
        *reinterpret_cast<shortx8*>(k_lds + lds_off32w(row, col * 2, NAT_RS, KSWZ)) = kv_;
        *reinterpret_cast<shortx8*>(v_lds + lds_off32w(row, col * 2, NAT_RS, KSWZ)) = vv;
        const int rot = tid & 7;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          *reinterpret_cast<short*>(kt_lds + lds_off32(col + i, row * 2, KT_RS)) = kv_[i];
        }
      }
    }
    __syncthreads();

    // ---- S^T = K Q^T and dP^T = V dO^T ------------------------------------
    floatx16 st[2], dp[2];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
#pragma unroll
      for (int r = 0; r < 16; ++r) { st[t][r] = 0.f; dp[t][r] = 0.f; }
#pragma unroll
      for (int ks = 0; ks < NKS; ++ks) {
        shortx8 kf = *reinterpret_cast<const shortx8*>(
            k_lds + lds_off32w(t * 32 + l32, (ks * 16 + hi * 8) * 2, NAT_RS, KSWZ));
        st[t] = mfma32_bf16(kf, qf[ks], st[t]);
        shortx8 vf = *reinterpret_cast<const shortx8*>(
            v_lds + lds_off32w(t * 32 + l32, (ks * 16 + hi * 8) * 2, NAT_RS, KSWZ));
        dp[t] = mfma32_bf16(vf, dof[ks], dp[t]);
      }
    }

    // ---- dS^T = P^T (dP^T o dropout - delta) * scale ----------------------
    if (MASKED && q_abs < Sq_e) {
      const short* mrow = maskg + (long long)b * m_sb + (long long)h * m_sh
                        + (long long)q_abs * m_sq;
#pragma unroll
      for (int t = 0; t < 2; ++t)
#pragma unroll
        for (int g = 0; g < 4; ++g) {
          long long kvb = kv0 + 32 * t + 8 * g + 4 * hi;
          if (kvb + 3 < Skv_e) {
            shortx4 mv = *reinterpret_cast<const shortx4*>(mrow + kvb);
#pragma unroll
            for (int j = 0; j < 4; ++j) st[t][4 * g + j] += bf2f(mv[j]);
          } else {
#pragma unroll
            for (int j = 0; j < 4; ++j)
              if (kvb + j < Skv_e) st[t][4 * g + j] += bf2f(mrow[kvb + j]);
          }
        }
    }
    const bool bnd = (kv0 + KB > Skv_e) || (CAUSAL && kv0 + KB > q_abs + cdelta);
    const float inv_keep = DROPOUT ? 1.f / (1.f - pdrop) : 1.f;
    const unsigned thr24 = DROPOUT ? (unsigned)(pdrop * 16777216.f) : 0u;
    const long long bh_ll = (long long)b * H + h;
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      if (DROPOUT) {
#pragma unroll
        for (int g = 0; g < 4; ++g) {
          long long kvb = kv0 + 32 * t + 8 * g + 4 * hi;
          long long elem = (bh_ll * Sq + (q_abs + qglob0)) * (long long)Skv + kvb;
          uintx4 rv = philox10(rseed, roffset + (unsigned long long)(elem >> 2));
#pragma unroll
          for (int j = 0; j < 4; ++j)
            dp[t][4 * g + j] = ((rv[j] >> 8) >= thr24)
                                   ? dp[t][4 * g + j] * inv_keep : 0.f;
        }
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float p = __expf(st[t][r] - lse_v);   // Q pre-scaled at load
        if (bnd) {
          int kv_abs = kv0 + 32 * t + (r & 3) + 8 * (r >> 2) + 4 * hi;
          if (kv_abs >= Skv_e || (CAUSAL && kv_abs > q_abs + cdelta)) p = 0.f;
        }
        st[t][r] = p * (dp[t][r] - delta_v) * scale;
      }
    }

    // ---- dQ += dS K (A-frags via static packs + permlane exchange) --------
#pragma unroll
    for (int ks = 0; ks < KB / 16; ++ks) {
      const int t = ks >> 1;
      const int kp = ks & 1;
      const int b0 = 8 * kp;
      int P0 = pack_bf2(st[t][b0 + 0], st[t][b0 + 1]);
      int P1 = pack_bf2(st[t][b0 + 2], st[t][b0 + 3]);
      int P2 = pack_bf2(st[t][b0 + 4], st[t][b0 + 5]);
      int P3 = pack_bf2(st[t][b0 + 6], st[t][b0 + 7]);
      int O1 = hi ? P2 : P0;
      int O2 = hi ? P3 : P1;
      int S1 = hi ? P0 : P2;
      int S2 = hi ? P1 : P3;
      intx2 ra = __builtin_amdgcn_permlane32_swap(S1, S2, false, false);
      intx2 rb = __builtin_amdgcn_permlane32_swap(S2, S1, false, false);
      int X1 = hi ? rb[0] : ra[1];
      int X2 = hi ? ra[0] : rb[1];
      intx4 paw;
      paw[0] = hi ? X1 : O1;
      paw[1] = hi ? X2 : O2;
      paw[2] = hi ? O1 : X1;
      paw[3] = hi ? O2 : X2;
      shortx8 pa = *reinterpret_cast<shortx8*>(&paw);
#pragma unroll
      for (int dt = 0; dt < NDT; ++dt) {
        shortx8 bf = *reinterpret_cast<const shortx8*>(
            kt_lds + lds_off32(dt * 32 + l32, (ks * 16 + hi * 8) * 2, KT_RS));
        dq_acc[dt] = mfma32_bf16(pa, bf, dq_acc[dt]);
      }
    }
    __syncthreads();
  }

  // ---- epilogue -----------------------------------------------------------
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int rq = (r & 3) + 8 * (r >> 2) + 4 * hi;
    int row = q0w + rq;
    if (row >= Sq_e) continue;
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt)
      dqg[dqbase + (long long)row * dq_ss + dt * 32 + l32] = f2bf(dq_acc[dt][r]);
  }
}

void flash_attn_bwd_dq32(const void* dout, const void* q, const void* k,
                         const void* v, const float* lse, const float* delta,
                         void* dq, int64_t b, int64_t h, int64_t sq, int64_t skv,
                         int64_t dh, float scale, bool causal,
                         const int64_t* qs, const int64_t* ks, const int64_t* dos,
                         const int64_t* dqs, const void* mask, const int64_t* ms,
                         float pdrop, uint64_t seed, uint64_t offset,
                         hipStream_t s) {
  dim3 grid((unsigned)(b * h), (unsigned)cdiv((int)sq, 128));
  dim3 blk(256);
  const bool masked = mask != nullptr;
  const bool dropped = pdrop > 0.f;
  static const int64_t zs[3] = {0, 0, 0};
  if (!ms) ms = zs;
#define FDQ1(D, C, M, P)                                                        \
  hipLaunchKernelGGL((fa_bwd_dq32_kernel<D, C, M, P>), grid, blk, 0, s,         \
                     (const short*)dout, (const short*)q, (const short*)k,      \
                     (const short*)v, lse, delta, (short*)dq, (int)b, (int)h,   \
                     (int)sq, (int)skv, scale, qs[0], qs[1], qs[2],             \
                     ks[0], ks[1], ks[2], dos[0], dos[1], dos[2],               \
                     dqs[0], dqs[1], dqs[2], (const short*)mask,                \
                     ms[0], ms[1], ms[2], pdrop, seed, offset)
#define FDQ(D, C)                                                               \
  do {                                                                          \
    if (masked && dropped) FDQ1(D, C, true, true);                              \
    else if (masked)       FDQ1(D, C, true, false);                             \
    else if (dropped)      FDQ1(D, C, false, true);                             \
    else                   FDQ1(D, C, false, false);                            \
  } while (0)
  if (dh == 128) { if (causal) FDQ(128, true); else FDQ(128, false); }
  else           { if (causal) FDQ(64, true);  else FDQ(64, false); }
#undef FDQ
#undef FDQ1
}

// ---------------------------------------------------------------------------
// backward dV / dK, 32x32 split variant (v3).
//
// Orientation: P = mfma(Q, K) and dP = mfma(dO, V) put the kv column in
// the LANE dim, so P / dS feed dV^T = dO^T P and dK^T = Q^T dS as
// B-fragments via the forward's static-pack + permlane32_swap exchange --
// no lane<->reg transpose, no LDS round trip (the v1 split's failure).
// Split into dV and dK kernels so each fits 2 waves/SIMD (the combined
// variant needs 128 accumulator VGPRs and runs at occupancy 1, measured
// slower than the 16x16 kernel).  Q / dO A-fragments are read straight
// from global (all 4 waves read the same rows -> L2-served); only the
// transposed operand (dO^T or Q^T) is staged in LDS, double-buffered.
// Outputs are accumulated transposed and written as packed 8 B stores.
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL, bool IS_DK, bool MASKED, bool DROPOUT,
          bool VARLEN = false>
__launch_bounds__(256, 2)
__global__ void fa_bwd_dkv32_kernel(const short* __restrict__ dog, const short* __restrict__ qg,
                                    const short* __restrict__ kg, const short* __restrict__ vg,
                                    const float* __restrict__ lseg, const float* __restrict__ deltag,
                                    short* __restrict__ outg,
                                    int B, int H, int Sq, int Skv, float scale,
                                    long long q_sb, long long q_sh, long long q_ss,
                                    long long k_sb, long long k_sh, long long k_ss,
                                    long long do_sb, long long do_sh, long long do_ss,
                                    long long dk_sb, long long dk_sh, long long dk_ss,
                                    const short* __restrict__ maskg, long long m_sb,
                                    long long m_sh, long long m_sq,
                                    float pdrop, unsigned long long rseed,
                                    unsigned long long roffset,
                                    const int* __restrict__ cu_q = nullptr,
                                    const int* __restrict__ cu_k = nullptr,
                                    const int* __restrict__ bmap = nullptr) {
  constexpr int NW = 4;
  constexpr int NT = NW * 64;
  constexpr int KVB = NW * 32;     // 128 kv rows / block
  constexpr int QT = 32;           // q tile (one 32-q mfma tile)
  constexpr int NKS = D / 16;
  constexpr int NDT = D / 32;
  constexpr unsigned TR_RS = 64 * 2;  // transposed rows padded to 64 cols
  // transposed operand (dO^T for dV, Q^T for dK): rows = d (128), padded to
  // 64 bf16 columns so lds_off32's 8-row XOR swizzle stays in-row
  __shared__ char t_lds[2][D * 64 * 2];
  __shared__ float stat_s[2][2 * QT];  // [lse | delta]

  const int kvblk = blockIdx.y;     // slow dim; kv0=0 (longest) dispatches first
  const int bh = blockIdx.x;
  int b = bh / H, h = bh % H;
  int kv0, Sq_e = Sq, Skv_e = Skv, cdelta = 0, qglob0 = 0;
  long long qbase, dobase, kvbase, outbase;
  if (VARLEN) {
    b = 0; h = bh;
    const int seq = bmap[2 * kvblk];
    kv0 = bmap[2 * kvblk + 1];
    const int qs0 = cu_q[seq], ks0 = cu_k[seq];
    Sq_e = cu_q[seq + 1] - qs0;
    Skv_e = cu_k[seq + 1] - ks0;
    cdelta = Skv_e - Sq_e;
    qglob0 = qs0;
    qbase = (long long)qs0 * q_ss + (long long)h * q_sh;
    dobase = (long long)qs0 * do_ss + (long long)h * do_sh;
    kvbase = (long long)ks0 * k_ss + (long long)h * k_sh;
    outbase = (long long)ks0 * dk_ss + (long long)h * dk_sh;
  } else {
    kv0 = kvblk * KVB;
    qbase = (long long)b * q_sb + (long long)h * q_sh;
    dobase = (long long)b * do_sb + (long long)h * do_sh;
    kvbase = (long long)b * k_sb + (long long)h * k_sh;
    outbase = (long long)b * dk_sb + (long long)h * dk_sh;
  }
  const long long lse_base = ((long long)bh) * Sq + qglob0;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wq = tid >> 6;
  const int l32 = lane & 31;
  const int hi = lane >> 5;
  const int kv0w = kv0 + wq * 32;
  const int kv_lane = kv0w + l32;

  // K (and V for dK) as B-fragments (lane = kv row), loaded once
  shortx8 kf[NKS], vf[IS_DK ? NKS : 1];
  {
    bool ok = kv_lane < Skv_e;
#pragma unroll
    for (int ks = 0; ks < NKS; ++ks) {
      if (ok) {
        kf[ks] = *reinterpret_cast<const shortx8*>(
            kg + kvbase + (long long)kv_lane * k_ss + ks * 16 + hi * 8);
        if (IS_DK)
          vf[ks] = *reinterpret_cast<const shortx8*>(
              vg + kvbase + (long long)kv_lane * k_ss + ks * 16 + hi * 8);
      } else {
        for (int i = 0; i < 8; ++i) { kf[ks][i] = 0; if (IS_DK) vf[ks][i] = 0; }
      }
    }
  }

  floatx16 acc[NDT];
#pragma unroll
  for (int dt = 0; dt < NDT; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc[dt][r] = 0.f;

  // stage the transposed operand tile (rotated scalar writes) + stats
  auto stage = [&](int buf, int q0) {
    if (tid < QT) {
      int qa = q0 + tid;
      stat_s[buf][tid] = (qa < Sq_e) ? lseg[lse_base + qa] : 1e30f;
      if (IS_DK)
        stat_s[buf][QT + tid] = (qa < Sq_e) ? deltag[lse_base + qa] : 0.f;
    }
    const int rot = tid & 7;
    for (int flat = tid * 8; flat < QT * D; flat += NT * 8) {
      int row = flat / D, col = flat % D;   // row = q, col = d
      shortx8 v;
      if (q0 + row < Sq_e) {
        const short* src = IS_DK
            ? qg + qbase + (long long)(q0 + row) * q_ss + col
            : dog + dobase + (long long)(q0 + row) * do_ss + col;
        v = *reinterpret_cast<const shortx8*>(src);
      } else {
        for (int i = 0; i < 8; ++i) v[i] = 0;
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int i = (j + rot) & 7;
        *reinterpret_cast<short*>(t_lds[buf] + lds_off32(col + i, row * 2, TR_RS)) = v[i];
      }
    }
  };

  const int q_start = CAUSAL ? (max(0, kv0 - cdelta) / QT) * QT : 0;
  stage(0, q_start);
  __syncthreads();
  int cur = 0;
  for (int q0 = q_start; q0 < Sq_e; q0 += QT) {
    if (q0 + QT < Sq_e) stage(cur ^ 1, q0 + QT);

    // Q (and dO for dK) A-fragments straight from global: lane = q row
    const int q_lane = q0 + l32;
    const bool qok = q_lane < Sq_e;
    floatx16 st, dp;
#pragma unroll
    for (int r = 0; r < 16; ++r) { st[r] = 0.f; dp[r] = 0.f; }
#pragma unroll
    for (int ks = 0; ks < NKS; ++ks) {
      shortx8 qa;
      if (qok)
        qa = *reinterpret_cast<const shortx8*>(
            qg + qbase + (long long)q_lane * q_ss + ks * 16 + hi * 8);
      else
        for (int i = 0; i < 8; ++i) qa[i] = 0;
      st = mfma32_bf16(qa, kf[ks], st);
      if (IS_DK) {
        shortx8 da;
        if (qok)
          da = *reinterpret_cast<const shortx8*>(
              dog + dobase + (long long)q_lane * do_ss + ks * 16 + hi * 8);
        else
          for (int i = 0; i < 8; ++i) da[i] = 0;
        dp = mfma32_bf16(da, vf[ks], dp);
      }
    }

    // per-reg q stats + exp/mask.  For dV, P = exp(S*scale - lse[q]); for
    // dK additionally dS = P (dP - delta[q]) * scale -- but P needs lse
    // there too, so dK stages BOTH: lse comes from a second pass of
    // broadcast loads straight from global (L2-hot after the dV kernel).
    const bool bnd = (q0 + QT > Sq_e) || (kv0w + 32 > Skv_e) ||
                     (CAUSAL && kv0w + 32 > q0 + cdelta);
    const float inv_keep = DROPOUT ? 1.f / (1.f - pdrop) : 1.f;
    const unsigned thr24 = DROPOUT ? (unsigned)(pdrop * 16777216.f) : 0u;
    const long long bh_ll = (long long)b * H + h;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int rq = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int q_abs = q0 + rq;
      float sc = st[r] * scale;
      if (MASKED && q_abs < Sq_e && kv_lane < Skv_e)
        sc += bf2f(maskg[(long long)b * m_sb + (long long)h * m_sh
                         + (long long)q_abs * m_sq + kv_lane]);
      float p = __expf(sc - stat_s[cur][rq]);
      if (bnd) {
        if (q_abs >= Sq_e || kv_lane >= Skv_e ||
            (CAUSAL && kv_lane > q_abs + cdelta)) p = 0.f;
      }
      bool keep = true;
      if (DROPOUT)
        keep = fa_keep(rseed, roffset, bh_ll, Sq, Skv, q_abs + qglob0,
                       kv_lane, thr24);
      if (IS_DK) {
        float dpe = dp[r];
        if (DROPOUT) dpe = keep ? dpe * inv_keep : 0.f;
        st[r] = p * (dpe - stat_s[cur][QT + rq]) * scale;
      } else {
        if (DROPOUT) p = keep ? p * inv_keep : 0.f;
        st[r] = p;
      }
    }

    // ---- acc += (dO^T P) or (Q^T dS): B-frags via pack+permlane ----------
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int b0 = 8 * ks;
      int P0 = pack_bf2(st[b0 + 0], st[b0 + 1]);
      int P1 = pack_bf2(st[b0 + 2], st[b0 + 3]);
      int P2 = pack_bf2(st[b0 + 4], st[b0 + 5]);
      int P3 = pack_bf2(st[b0 + 6], st[b0 + 7]);
      int O1 = hi ? P2 : P0, O2 = hi ? P3 : P1;
      int S1 = hi ? P0 : P2, S2 = hi ? P1 : P3;
      intx2 ra = __builtin_amdgcn_permlane32_swap(S1, S2, false, false);
      intx2 rb = __builtin_amdgcn_permlane32_swap(S2, S1, false, false);
      int X1 = hi ? rb[0] : ra[1], X2 = hi ? ra[0] : rb[1];
      intx4 paw;
      paw[0] = hi ? X1 : O1; paw[1] = hi ? X2 : O2;
      paw[2] = hi ? O1 : X1; paw[3] = hi ? O2 : X2;
      shortx8 frag = *reinterpret_cast<shortx8*>(&paw);
#pragma unroll
      for (int dt = 0; dt < NDT; ++dt) {
        shortx8 tfr = *reinterpret_cast<const shortx8*>(
            t_lds[cur] + lds_off32(dt * 32 + l32, (ks * 16 + hi * 8) * 2, TR_RS));
        acc[dt] = mfma32_bf16(tfr, frag, acc[dt]);
      }
    }
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: transposed accumulators -> natural rows ------------------
  if (kv_lane < Skv_e) {
    const long long orow = outbase + (long long)kv_lane * dk_ss;
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt)
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        shortx4 pv;
#pragma unroll
        for (int i = 0; i < 4; ++i) pv[i] = f2bf(acc[dt][g * 4 + i]);
        int d = dt * 32 + 8 * g + 4 * hi;
        *reinterpret_cast<shortx4*>(outg + orow + d) = pv;
      }
  }
}

void flash_attn_bwd_dkv32(const void* dout, const void* q, const void* k,
                          const void* v, const float* lse, const float* delta,
                          void* dk, void* dv, int64_t b, int64_t h, int64_t sq,
                          int64_t skv, int64_t dh, float scale, bool causal,
                          const int64_t* qs, const int64_t* ks, const int64_t* dos,
                          const int64_t* dks, const void* mask, const int64_t* ms,
                          float pdrop, uint64_t seed, uint64_t offset,
                          hipStream_t s) {
  dim3 grid((unsigned)(b * h), (unsigned)cdiv((int)skv, 128));
  dim3 blk(256);
  const bool masked = mask != nullptr;
  const bool dropped = pdrop > 0.f;
  static const int64_t zs[3] = {0, 0, 0};
  if (!ms) ms = zs;
#define FDKV2(D, C, DK, M, P, OUT)                                             \
  hipLaunchKernelGGL((fa_bwd_dkv32_kernel<D, C, DK, M, P>), grid, blk, 0, s,   \
                     (const short*)dout, (const short*)q, (const short*)k,     \
                     (const short*)v, lse, delta, (short*)(OUT),               \
                     (int)b, (int)h, (int)sq, (int)skv, scale,                 \
                     qs[0], qs[1], qs[2], ks[0], ks[1], ks[2],                 \
                     dos[0], dos[1], dos[2], dks[0], dks[1], dks[2],           \
                     (const short*)mask, ms[0], ms[1], ms[2], pdrop, seed, offset)
#define FDKV1(D, C, M, P)                                                      \
  do { FDKV2(D, C, false, M, P, dv); FDKV2(D, C, true, M, P, dk); } while (0)
#define FDKV(D, C)                                                             \
  do {                                                                         \
    if (masked && dropped) FDKV1(D, C, true, true);                            \
    else if (masked)       FDKV1(D, C, true, false);                           \
    else if (dropped)      FDKV1(D, C, false, true);                           \
    else                   FDKV1(D, C, false, false);                          \
  } while (0)
  if (dh == 128) { if (causal) FDKV(128, true); else FDKV(128, false); }
  else           { if (causal) FDKV(64, true);  else FDKV(64, false); }
#undef FDKV
#undef FDKV1
#undef FDKV2
}

// ---------------------------------------------------------------------------
// varlen (ragged, flash_attn_unpadded) launchers: ONE launch for the whole
// packed batch; per-block sequence bounds from bmap/cu_seqlens.
// Packed layout [total, H, D] contiguous; lse/delta [H, total_q].
// ---------------------------------------------------------------------------
void flash_attn_varlen_fwd32(const void* q, const void* k, const void* v,
                             void* o, float* lse, int64_t h, int64_t hkv,
                             int64_t total_q, int64_t total_k, int64_t dh,
                             float scale, bool causal, int64_t nblocks,
                             const int* cu_q, const int* cu_k, const int* bmap,
                             float pdrop, uint64_t seed, uint64_t offset,
                             hipStream_t s) {
  dim3 grid((unsigned)h, (unsigned)nblocks);
  dim3 blk(256);
  const bool dropped = pdrop > 0.f;
  const long long q_ss = h * dh, k_ss = hkv * dh;
#define VFF(D, C, P)                                                           \
  hipLaunchKernelGGL((fa_fwd32_kernel<D, C, false, P, true>), grid, blk, 0, s, \
                     (const short*)q, (const short*)k, (const short*)v,        \
                     (short*)o, lse, 1, (int)h, (int)hkv, (int)total_q,        \
                     (int)total_k, scale, 0, dh, q_ss, 0, dh, k_ss,            \
                     0, dh, q_ss, nullptr, 0, 0, 0, pdrop, seed, offset,       \
                     cu_q, cu_k, bmap)
#define VF(D, C) do { if (dropped) VFF(D, C, true); else VFF(D, C, false); } while (0)
  if (dh == 128) { if (causal) VF(128, true); else VF(128, false); }
  else           { if (causal) VF(64, true);  else VF(64, false); }
#undef VF
#undef VFF
}

void flash_attn_varlen_bwd32(const void* dout, const void* q, const void* k,
                             const void* v, const float* lse, const float* delta,
                             void* dq, void* dk, void* dv, int64_t h,
                             int64_t total_q, int64_t total_k, int64_t dh,
                             float scale, bool causal, int64_t nqblocks,
                             int64_t nkvblocks, const int* cu_q, const int* cu_k,
                             const int* qbmap, const int* kvbmap, float pdrop,
                             uint64_t seed, uint64_t offset, hipStream_t s) {
  dim3 blk(256);
  const bool dropped = pdrop > 0.f;
  const long long ss = h * dh;
#define VDQ(D, C, P)                                                           \
  hipLaunchKernelGGL((fa_bwd_dq32_kernel<D, C, false, P, true>),               \
                     dim3((unsigned)h, (unsigned)nqblocks), blk, 0, s,         \
                     (const short*)dout, (const short*)q, (const short*)k,     \
                     (const short*)v, lse, delta, (short*)dq, 1, (int)h,       \
                     (int)total_q, (int)total_k, scale, 0, dh, ss, 0, dh, ss,  \
                     0, dh, ss, 0, dh, ss, nullptr, 0, 0, 0, pdrop, seed,      \
                     offset, cu_q, cu_k, qbmap)
#define VDKV(D, C, DK, P, OUT)                                                 \
  hipLaunchKernelGGL((fa_bwd_dkv32_kernel<D, C, DK, false, P, true>),          \
                     dim3((unsigned)h, (unsigned)nkvblocks), blk, 0, s,        \
                     (const short*)dout, (const short*)q, (const short*)k,     \
                     (const short*)v, lse, delta, (short*)(OUT), 1, (int)h,    \
                     (int)total_q, (int)total_k, scale, 0, dh, ss, 0, dh, ss,  \
                     0, dh, ss, 0, dh, ss, nullptr, 0, 0, 0, pdrop, seed,      \
                     offset, cu_q, cu_k, kvbmap)
#define VB(D, C, P)                                                            \
  do { VDQ(D, C, P); VDKV(D, C, false, P, dv); VDKV(D, C, true, P, dk); } while (0)
#define VBD(D, C) do { if (dropped) VB(D, C, true); else VB(D, C, false); } while (0)
  if (dh == 128) { if (causal) VBD(128, true); else VBD(128, false); }
  else           { if (causal) VBD(64, true);  else VBD(64, false); }
#undef VBD
#undef VB
#undef VDKV
#undef VDQ
}

// debug/test utility: materialize the attention-dropout keep mask the FA
// kernels derive from (seed, offset) -- lets tests compare fwd/bwd against
// an exact CPU oracle using the same mask.
__global__ void fa_dropout_mask_kernel(unsigned char* __restrict__ out,
                                       float p, unsigned long long seed,
                                       unsigned long long offset,
                                       long long total) {
  long long i4 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long base = i4 * 4;
  if (base >= total) return;
  unsigned thr24 = (unsigned)(p * 16777216.f);
  uintx4 rv = philox10(seed, offset + (unsigned long long)i4);
#pragma unroll
  for (int j = 0; j < 4; ++j)
    if (base + j < total) out[base + j] = ((rv[j] >> 8) >= thr24) ? 1 : 0;
}

void fa_dropout_mask(void* out, int64_t total, float p, uint64_t seed,
                     uint64_t offset, hipStream_t s) {
  long long n4 = (total + 3) / 4;
  dim3 g((unsigned)hmin<long long>((n4 + 255) / 256, 1 << 30));
  hipLaunchKernelGGL(fa_dropout_mask_kernel, g, dim3(256), 0, s,
                     (unsigned char*)out, p, seed, offset, total);
}

}  // namespace pa
