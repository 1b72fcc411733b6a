// FlashAttention-2 forward + backward for gfx950 (CDNA4), bf16, head_dim
// 64/128, causal or full, MHA/GQA (GQA fwd; bwd requires hkv==h for the
// in-kernel path -- the Python layer repeats KV otherwise).
//
// MI355X-first design (NOT a port of the reference's dynloaded
// libflashattn -- paddle/phi/kernels/gpu/flash_attn_kernel.cu:41 is a
// dispatch shim; the math here is re-derived):
//   * MFMA 16x16x32 bf16 tiles; 4 waves/block; each wave owns 16 rows of
//     the 64-row Q (fwd) / KV (bwd-dkv) tile.
//   * K / V^T staged in LDS with the XOR-16B swizzle (guide §6 G4) so
//     column-sliced ds_read_b128 is ~2-way conflict free.
//   * online softmax entirely in registers; row stats shared across the
//     16-lane C-tile groups via shfl_xor(1,2,4,8).
//   * LSE saved (fp32) for backward + ring/context-parallel merges.
//   * backward = 3 kernels (delta, dKV, dQ) -- no atomics, S recomputed
//     from Q,K,LSE per FA2.
#include "common.h"
#include "api.h"

namespace pa {

#define LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")

// lds byte offset helpers (row stride in bytes RS must be multiple of 32)
__device__ __forceinline__ unsigned lds_off(unsigned row, unsigned col_bytes,
                                            unsigned row_stride) {
  return row * row_stride + (col_bytes ^ ((row & 7u) << 4));
}

// read a 16B swizzled fragment (8 bf16) from LDS
__device__ __forceinline__ shortx8 lds_read8(const char* lds, unsigned row,
                                             unsigned col_elem, unsigned row_stride) {
  return *reinterpret_cast<const shortx8*>(lds + lds_off(row, col_elem * 2, row_stride));
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL>
__launch_bounds__(512)
__global__ void fa_fwd_kernel(const short* __restrict__ qg, const short* __restrict__ kg,
                              const short* __restrict__ vg, short* __restrict__ og,
                              float* __restrict__ lseg, int B, int H, int HKV,
                              int Sq, int Skv, float scale,
                              long long q_sb, long long q_sh, long long q_ss,
                              long long k_sb, long long k_sh, long long k_ss,
                              long long o_sb, long long o_sh, long long o_ss) {
  constexpr int NW = 8;         // waves per block (16 q rows each)
  constexpr int QB = 16 * NW, KB = 64;
  constexpr int NT = NW * 64;   // threads
  constexpr int NKK = D / 32;   // mfma k-steps over head dim
  constexpr int NDT = D / 16;   // output d tiles
  // LDS layout
  constexpr unsigned K_RS = D * 2;            // K tile row stride bytes
  constexpr unsigned VT_RS = KB * 2;          // V^T tile row stride
  __shared__ char k_lds[KB * D * 2];
  __shared__ char vt_lds[D * KB * 2];
  __shared__ char p_lds[NW * 16 * KB * 2];    // per-wave P tiles

  const int qblk = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int hkv = h / (H / HKV);
  const int q0 = qblk * QB;
  const long long qbase = (long long)b * q_sb + (long long)h * q_sh;
  const long long kbase = (long long)b * k_sb + (long long)hkv * k_sh;
  const long long obase = (long long)b * o_sb + (long long)h * o_sh;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wq = tid >> 6;          // wave id = which 16 q-rows
  const int l16 = lane & 15;        // col within C tile
  const int lg = lane >> 4;         // 16-lane group id (0..3)

  // Q fragments: A[i=row16][k] ; lane holds row l16, k = lg*8.. per kk-step
  shortx8 qf[NKK];
  {
    int row = q0 + wq * 16 + l16;
    bool ok = row < Sq;
#pragma unroll
    for (int kk = 0; kk < NKK; ++kk) {
      if (ok)
        qf[kk] = *reinterpret_cast<const shortx8*>(qg + qbase + (long long)row * q_ss + kk * 32 + lg * 8);
      else
        for (int i = 0; i < 8; ++i) qf[kk][i] = 0;
    }
  }

  floatx4 oacc[NDT];
#pragma unroll
  for (int dt = 0; dt < NDT; ++dt) oacc[dt] = {0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }

  const int kv_end = CAUSAL ? min(Skv, q0 + QB) : Skv;

  // K staging source offsets (pre-swizzled global address so the async
  // global_load_lds DMA lands the swizzled layout into linear LDS -- guide
  // §5 m173 pattern; the read side XORs the same mask back out)
  int k_row[KB * D / (NT * 8)], k_colp[KB * D / (NT * 8)];
#pragma unroll
  for (int it = 0; it < KB * D / (NT * 8); ++it) {
    int flat = it * NT * 8 + tid * 8;
    int row = flat / D, col = flat % D;
    k_row[it] = row;
    k_colp[it] = col ^ ((row & 7) << 3);
  }

  for (int kv0 = 0; kv0 < kv_end; kv0 += KB) {
    // ---- stage K tile [KB][D] (swizzled rows) -----------------------------
    if (kv0 + KB <= Skv) {
      // fast path: async DMA straight to LDS, no VGPR round-trip
#pragma unroll
      for (int it = 0; it < KB * D / (NT * 8); ++it) {
        const short* src = kg + kbase + (long long)(kv0 + k_row[it]) * k_ss + k_colp[it];
        char* dst = k_lds + it * NT * 16 + (tid >> 6) * 64 * 16;
        __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) unsigned int*)src,
                                         (__attribute__((address_space(3))) unsigned int*)dst,
                                         16, 0, 0);
      }
    } else {
      constexpr int elems = KB * D;
      for (int flat = tid * 8; flat < elems; flat += NT * 8) {
        int row = flat / D, col = flat % D;
        shortx8 val;
        if (kv0 + row < Skv)
          val = *reinterpret_cast<const shortx8*>(kg + kbase + (long long)(kv0 + row) * k_ss + col);
        else
          for (int i = 0; i < 8; ++i) val[i] = 0;
        *reinterpret_cast<shortx8*>(k_lds + lds_off(row, col * 2, K_RS)) = val;
      }
    }
    {
      constexpr int elems = KB * D;
      // ---- stage V^T tile [D][KB] ------------------------------------------
      for (int flat = tid * 8; flat < elems; flat += NT * 8) {
        int row = flat / D, col = flat % D;  // row=kv, col=d
        shortx8 val;
        if (kv0 + row < Skv)
          val = *reinterpret_cast<const shortx8*>(vg + kbase + (long long)(kv0 + row) * k_ss + col);
        else
          for (int i = 0; i < 8; ++i) val[i] = 0;
        // rotate write order by lane: without it all 16 lanes of a row-
        // group hit ONE bank (16-way conflict, the PMC hotspot); rotated
        // they spread over 8 banks (2-way).
        const int rot = threadIdx.x & 7;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          *reinterpret_cast<short*>(vt_lds + lds_off(col + i, row * 2, VT_RS)) = val[i];
        }
      }
    }
    __syncthreads();

    // ---- S = Q K^T --------------------------------------------------------
    floatx4 sacc[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      sacc[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < NKK; ++kk) {
        shortx8 kf = lds_read8(k_lds, nt * 16 + l16, kk * 32 + lg * 8, K_RS);
        sacc[nt] = mfma_bf16(qf[kk], kf, sacc[nt]);
      }
    }

    // ---- mask + scale + online softmax ------------------------------------
    const int q_abs = q0 + wq * 16 + lg * 4;  // + r
    const bool boundary = (kv0 + KB > Skv) || (CAUSAL && kv0 + KB > q_abs);
    float mx[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
    if (boundary) {
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int kv_abs = kv0 + nt * 16 + l16;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float val = sacc[nt][r] * scale;
          if (kv_abs >= Skv || (CAUSAL && kv_abs > q_abs + r)) val = -INFINITY;
          sacc[nt][r] = val;
          mx[r] = fmaxf(mx[r], val);
        }
      }
    } else {
#pragma unroll
      for (int nt = 0; nt < 4; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float val = sacc[nt][r] * scale;
          sacc[nt][r] = val;
          mx[r] = fmaxf(mx[r], val);
        }
    }
    float psum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      mx[r] = group16_reduce(mx[r], MaxOp());
      // defer-max (guide T13): skip the O(NDT) rescale while the running
      // max hasn't grown by more than THR (P stays bounded by e^THR)
      const float THR = 8.f;
      if (mx[r] > m_run[r] + THR || m_run[r] == -INFINITY) {
        float m_new = fmaxf(m_run[r], mx[r]);
        float corr = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - m_new);
        l_run[r] *= corr;
#pragma unroll
        for (int dt = 0; dt < NDT; ++dt) oacc[dt][r] *= corr;
        m_run[r] = m_new;
      }
      psum[r] = 0.f;
    }
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = (sacc[nt][r] == -INFINITY) ? 0.f : __expf(sacc[nt][r] - m_run[r]);
        sacc[nt][r] = p;
        psum[r] += p;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      psum[r] = group16_reduce(psum[r], SumOp());
      l_run[r] += psum[r];
    }

    // ---- P -> per-wave LDS (bf16, swizzled 128B rows) ---------------------
    char* pw = p_lds + wq * (16 * KB * 2);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        unsigned row = lg * 4 + r, col = nt * 16 + l16;
        *reinterpret_cast<short*>(pw + lds_off(row, col * 2, KB * 2)) = f2bf(sacc[nt][r]);
      }
    }
    LGKM0();

    // ---- O += P V ---------------------------------------------------------
#pragma unroll
    for (int kk = 0; kk < KB / 32; ++kk) {
      shortx8 pf = lds_read8(pw, l16, kk * 32 + lg * 8, KB * 2);
#pragma unroll
      for (int dt = 0; dt < NDT; ++dt) {
        shortx8 vf = lds_read8(vt_lds, dt * 16 + l16, kk * 32 + lg * 8, VT_RS);
        oacc[dt] = mfma_bf16(pf, vf, oacc[dt]);
      }
    }
    __syncthreads();
  }

  // ---- finalize: O /= l ; write O + LSE -----------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = q0 + wq * 16 + lg * 4 + r;
    if (row >= Sq) continue;
    float inv_l = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt)
      og[obase + (long long)row * o_ss + dt * 16 + l16] = f2bf(oacc[dt][r] * inv_l);
    if (l16 == 0)
      lseg[(long long)(b * H + h) * Sq + row] =
          (l_run[r] > 0.f) ? m_run[r] + __logf(l_run[r]) : -INFINITY;
  }
}

// ---------------------------------------------------------------------------
// backward: delta[row] = sum_d dO*O
// ---------------------------------------------------------------------------
template <int D>
__global__ void fa_bwd_delta_kernel(const short* __restrict__ dog, const short* __restrict__ og,
                                    float* __restrict__ delta, int H, int Sq,
                                    long long rows,
                                    long long do_sb, long long do_sh, long long do_ss,
                                    long long o_sb, long long o_sh, long long o_ss) {
  long long wid = ((long long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int lane = threadIdx.x & 63;
  long long nwaves = ((long long)gridDim.x * blockDim.x) >> 6;
  for (long long r = wid; r < rows; r += nwaves) {
    int sq = (int)(r % Sq);
    int h = (int)((r / Sq) % H);
    int b = (int)(r / ((long long)Sq * H));
    const long long dob = b * do_sb + h * do_sh + sq * do_ss;
    const long long ob = b * o_sb + h * o_sh + sq * o_ss;
    float acc = 0.f;
    for (int j = lane * 8; j < D; j += 64 * 8) {
      shortx8 a = *reinterpret_cast<const shortx8*>(dog + dob + j);
      shortx8 bb = *reinterpret_cast<const shortx8*>(og + ob + j);
#pragma unroll
      for (int k = 0; k < 8; ++k) acc += bf2f(a[k]) * bf2f(bb[k]);
    }
    acc = wave_reduce(acc, SumOp());
    if (lane == 0) delta[r] = acc;
  }
}

// ---------------------------------------------------------------------------
// backward dK/dV: block owns a 64-row KV tile; waves own 16 kv rows each;
// iterate q tiles.  S^T computed as K·Q^T so kv is the C-row.
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL>
__launch_bounds__(512)
__global__ void fa_bwd_dkv_kernel(const short* __restrict__ dog, const short* __restrict__ qg,
                                  const short* __restrict__ kg, const short* __restrict__ vg,
                                  const float* __restrict__ lseg, const float* __restrict__ deltag,
                                  short* __restrict__ dkg, short* __restrict__ dvg,
                                  int B, int H, int Sq, int Skv, float scale,
                                  long long q_sb, long long q_sh, long long q_ss,
                                  long long k_sb, long long k_sh, long long k_ss,
                                  long long do_sb, long long do_sh, long long do_ss,
                                  long long dk_sb, long long dk_sh, long long dk_ss) {
  constexpr int NW = 8;               // waves; block owns NW*16 kv rows
  constexpr int NT = NW * 64;
  constexpr int KVEXT = NW * 16;
  constexpr int QB = 64;
  constexpr int NKK = D / 32;
  constexpr int NDT = D / 16;
  constexpr unsigned NAT_RS = D * 2;   // natural [64][D]
  constexpr unsigned TR_RS = QB * 2;   // transposed [D][64]
  __shared__ char q_lds[QB * D * 2];
  __shared__ char qt_lds[D * QB * 2];
  __shared__ char do_lds[QB * D * 2];
  __shared__ char dot_lds[D * QB * 2];
  __shared__ char ps_lds[NW * 16 * QB * 2];

  const int kvblk = blockIdx.y;     // slow dim; kv0=0 (longest) first
  const int bh = blockIdx.x;
  const int kv0 = kvblk * KVEXT;
  const int b = bh / H, h = bh % H;
  const long long qbase = (long long)b * q_sb + (long long)h * q_sh;
  const long long dobase = (long long)b * do_sb + (long long)h * do_sh;
  const long long kvbase = (long long)b * k_sb + (long long)h * k_sh;
  const long long dkbase = (long long)b * dk_sb + (long long)h * dk_sh;
  const long long lse_base = ((long long)bh) * Sq;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wq = tid >> 6;
  const int l16 = lane & 15;
  const int lg = lane >> 4;

  // K,V fragments (A-layout rows = kv)
  shortx8 kf[NKK], vf[NKK];
  {
    int row = kv0 + wq * 16 + l16;
    bool ok = row < Skv;
#pragma unroll
    for (int kk = 0; kk < NKK; ++kk) {
      if (ok) {
        kf[kk] = *reinterpret_cast<const shortx8*>(kg + kvbase + (long long)row * k_ss + kk * 32 + lg * 8);
        vf[kk] = *reinterpret_cast<const shortx8*>(vg + kvbase + (long long)row * k_ss + kk * 32 + lg * 8);
      } else {
        for (int i = 0; i < 8; ++i) { kf[kk][i] = 0; vf[kk][i] = 0; }
      }
    }
  }

  floatx4 dk_acc[NDT], dv_acc[NDT];
#pragma unroll
  for (int dt = 0; dt < NDT; ++dt) {
    dk_acc[dt] = {0.f, 0.f, 0.f, 0.f};
    dv_acc[dt] = {0.f, 0.f, 0.f, 0.f};
  }

  const int q_start = CAUSAL ? (kv0 / QB) * QB : 0;

  // DMA staging maps for the natural [64][D] tiles (pre-swizzled source)
  int n_row[QB * D / (NT * 8)], n_colp[QB * D / (NT * 8)];
#pragma unroll
  for (int it = 0; it < QB * D / (NT * 8); ++it) {
    int flat = it * NT * 8 + tid * 8;
    int row = flat / D, col = flat % D;
    n_row[it] = row;
    n_colp[it] = col ^ ((row & 7) << 3);
  }

  for (int q0 = q_start; q0 < Sq; q0 += QB) {
    // stage Q, dO natural (async DMA when in-bounds) + transposed (reg)
    if (q0 + QB <= Sq) {
#pragma unroll
      for (int it = 0; it < QB * D / (NT * 8); ++it) {
        const short* qsrc = qg + qbase + (long long)(q0 + n_row[it]) * q_ss + n_colp[it];
        const short* dsrc = dog + dobase + (long long)(q0 + n_row[it]) * do_ss + n_colp[it];
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)qsrc,
            (__attribute__((address_space(3))) unsigned int*)(q_lds + it * NT * 16 + (tid >> 6) * 64 * 16),
            16, 0, 0);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)dsrc,
            (__attribute__((address_space(3))) unsigned int*)(do_lds + it * NT * 16 + (tid >> 6) * 64 * 16),
            16, 0, 0);
      }
      // transposed copies from the DMA-landed natural tiles (LDS round
      // trip) -- re-reading Q/dO from global doubled this kernel's HBM
      // traffic, and the kernel is Q/dO-stream-bound
      __syncthreads();
      constexpr int elems = QB * D;
      const int rot = threadIdx.x & 7;
      for (int flat = tid * 8; flat < elems; flat += NT * 8) {
        int row = flat / D, col = flat % D;
        shortx8 qv = lds_read8(q_lds, row, col, NAT_RS);
        shortx8 dv = lds_read8(do_lds, row, col, NAT_RS);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          *reinterpret_cast<short*>(qt_lds + lds_off(col + i, row * 2, TR_RS)) = qv[i];
          *reinterpret_cast<short*>(dot_lds + lds_off(col + i, row * 2, TR_RS)) = dv[i];
        }
      }
    } else {
      constexpr int elems = QB * D;
      for (int flat = tid * 8; flat < elems; flat += NT * 8) {
        int row = flat / D, col = flat % D;
        shortx8 qv, dv;
        if (q0 + row < Sq) {
          qv = *reinterpret_cast<const shortx8*>(qg + qbase + (long long)(q0 + row) * q_ss + col);
          dv = *reinterpret_cast<const shortx8*>(dog + dobase + (long long)(q0 + row) * do_ss + col);
        } else {
          for (int i = 0; i < 8; ++i) { qv[i] = 0; dv[i] = 0; }
        }
        *reinterpret_cast<shortx8*>(q_lds + lds_off(row, col * 2, NAT_RS)) = qv;
        *reinterpret_cast<shortx8*>(do_lds + lds_off(row, col * 2, NAT_RS)) = dv;
        const int rot = threadIdx.x & 7;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          *reinterpret_cast<short*>(qt_lds + lds_off(col + i, row * 2, TR_RS)) = qv[i];
          *reinterpret_cast<short*>(dot_lds + lds_off(col + i, row * 2, TR_RS)) = dv[i];
        }
      }
    }
    __syncthreads();

    // per-lane lse/delta for the 4 q-subtiles (indexed by C-col = q)
    float lse_v[4], delta_v[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      int q_abs = q0 + nt * 16 + l16;
      lse_v[nt] = (q_abs < Sq) ? lseg[lse_base + q_abs] : INFINITY;
      delta_v[nt] = (q_abs < Sq) ? deltag[lse_base + q_abs] : 0.f;
    }

    // S^T = K Q^T ; P^T = exp(S^T*scale - lse[q])
    floatx4 pt[4];
    const int kv_abs0 = kv0 + wq * 16 + lg * 4;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      floatx4 st = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < NKK; ++kk) {
        shortx8 qfr = lds_read8(q_lds, nt * 16 + l16, kk * 32 + lg * 8, NAT_RS);
        st = mfma_bf16(kf[kk], qfr, st);
      }
      const int q_abs = q0 + nt * 16 + l16;
      const bool bnd = (q0 + QB > Sq) || (kv0 + KVEXT > Skv) ||
                       (CAUSAL && q0 < kv0 + KVEXT);
      if (bnd) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int kv_abs = kv_abs0 + r;
          float p = __expf(st[r] * scale - lse_v[nt]);
          if (q_abs >= Sq || kv_abs >= Skv || (CAUSAL && kv_abs > q_abs)) p = 0.f;
          pt[nt][r] = p;
        }
      } else {
#pragma unroll
        for (int r = 0; r < 4; ++r)
          pt[nt][r] = __expf(st[r] * scale - lse_v[nt]);
      }
    }

    // write P^T to wave LDS  [kv16][q64]
    char* pw = ps_lds + wq * (16 * QB * 2);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        *reinterpret_cast<short*>(pw + lds_off(lg * 4 + r, (nt * 16 + l16) * 2, QB * 2)) =
            f2bf(pt[nt][r]);
    LGKM0();

    // dV += P^T dO   (A = P^T from LDS, B = dO^T tile)
#pragma unroll
    for (int kk = 0; kk < QB / 32; ++kk) {
      shortx8 pa = lds_read8(pw, l16, kk * 32 + lg * 8, QB * 2);
#pragma unroll
      for (int dt = 0; dt < NDT; ++dt) {
        shortx8 bfr = lds_read8(dot_lds, dt * 16 + l16, kk * 32 + lg * 8, TR_RS);
        dv_acc[dt] = mfma_bf16(pa, bfr, dv_acc[dt]);
      }
    }

    // dP^T = V dO^T
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      floatx4 dp = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < NKK; ++kk) {
        shortx8 bfr = lds_read8(do_lds, nt * 16 + l16, kk * 32 + lg * 8, NAT_RS);
        dp = mfma_bf16(vf[kk], bfr, dp);
      }
      // dS^T = P^T * (dP^T - delta[q]) * scale
#pragma unroll
      for (int r = 0; r < 4; ++r)
        pt[nt][r] = pt[nt][r] * (dp[r] - delta_v[nt]) * scale;
    }
    LGKM0();  // ensure dV reads of pw done before overwrite
#pragma unroll
    for (int nt = 0; nt < 4; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        *reinterpret_cast<short*>(pw + lds_off(lg * 4 + r, (nt * 16 + l16) * 2, QB * 2)) =
            f2bf(pt[nt][r]);
    LGKM0();

    // dK += dS^T Q  (B = Q^T tile)
#pragma unroll
    for (int kk = 0; kk < QB / 32; ++kk) {
      shortx8 pa = lds_read8(pw, l16, kk * 32 + lg * 8, QB * 2);
#pragma unroll
      for (int dt = 0; dt < NDT; ++dt) {
        shortx8 bfr = lds_read8(qt_lds, dt * 16 + l16, kk * 32 + lg * 8, TR_RS);
        dk_acc[dt] = mfma_bf16(pa, bfr, dk_acc[dt]);
      }
    }
    __syncthreads();
  }

  // write dK, dV
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = kv0 + wq * 16 + lg * 4 + r;
    if (row >= Skv) continue;
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt) {
      dkg[dkbase + (long long)row * dk_ss + dt * 16 + l16] = f2bf(dk_acc[dt][r]);
      dvg[dkbase + (long long)row * dk_ss + dt * 16 + l16] = f2bf(dv_acc[dt][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// backward dQ: block owns a 64-row Q tile; iterate kv tiles.
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL>
__launch_bounds__(512)
__global__ void fa_bwd_dq_kernel(const short* __restrict__ dog, const short* __restrict__ qg,
                                 const short* __restrict__ kg, const short* __restrict__ vg,
                                 const float* __restrict__ lseg, const float* __restrict__ deltag,
                                 short* __restrict__ dqg, int B, int H, int Sq, int Skv,
                                 float scale,
                                 long long q_sb, long long q_sh, long long q_ss,
                                 long long k_sb, long long k_sh, long long k_ss,
                                 long long do_sb, long long do_sh, long long do_ss,
                                 long long dq_sb, long long dq_sh, long long dq_ss) {
  constexpr int NW = 8;               // waves; block owns NW*16 q rows
  constexpr int NT = NW * 64;
  constexpr int QEXT = NW * 16;
  constexpr int KB = 64;
  constexpr int NKK = D / 32;
  constexpr int NDT = D / 16;
  constexpr unsigned NAT_RS = D * 2;
  constexpr unsigned TR_RS = KB * 2;
  __shared__ char k_lds[KB * D * 2];
  __shared__ char v_lds[KB * D * 2];
  __shared__ char kt_lds[D * KB * 2];
  __shared__ char ps_lds[NW * 16 * KB * 2];

  const int qblk = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q0 = qblk * QEXT;
  const long long qbase = (long long)b * q_sb + (long long)h * q_sh;
  const long long dobase = (long long)b * do_sb + (long long)h * do_sh;
  const long long dqbase = (long long)b * dq_sb + (long long)h * dq_sh;
  const long long kvbase = (long long)b * k_sb + (long long)h * k_sh;
  const long long lse_base = ((long long)bh) * Sq;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wq = tid >> 6;
  const int l16 = lane & 15;
  const int lg = lane >> 4;

  shortx8 qf[NKK], dof[NKK];
  {
    int row = q0 + wq * 16 + l16;
    bool ok = row < Sq;
#pragma unroll
    for (int kk = 0; kk < NKK; ++kk) {
      if (ok) {
        qf[kk] = *reinterpret_cast<const shortx8*>(qg + qbase + (long long)row * q_ss + kk * 32 + lg * 8);
        dof[kk] = *reinterpret_cast<const shortx8*>(dog + dobase + (long long)row * do_ss + kk * 32 + lg * 8);
      } else {
        for (int i = 0; i < 8; ++i) { qf[kk][i] = 0; dof[kk][i] = 0; }
      }
    }
  }
  float lse_r[4], delta_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = q0 + wq * 16 + lg * 4 + r;
    lse_r[r] = (row < Sq) ? lseg[lse_base + row] : INFINITY;
    delta_r[r] = (row < Sq) ? deltag[lse_base + row] : 0.f;
  }

  floatx4 dq_acc[NDT];
#pragma unroll
  for (int dt = 0; dt < NDT; ++dt) dq_acc[dt] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = CAUSAL ? min(Skv, q0 + QEXT) : Skv;
  int n_row[KB * D / (NT * 8)], n_colp[KB * D / (NT * 8)];
#pragma unroll
  for (int it = 0; it < KB * D / (NT * 8); ++it) {
    int flat = it * NT * 8 + tid * 8;
    int row = flat / D, col = flat % D;
    n_row[it] = row;
    n_colp[it] = col ^ ((row & 7) << 3);
  }
  for (int kv0 = 0; kv0 < kv_end; kv0 += KB) {
    if (kv0 + KB <= Skv) {
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
      constexpr int elems = KB * D;
      const int rot = threadIdx.x & 7;
      for (int flat = tid * 8; flat < elems; flat += NT * 8) {
        int row = flat / D, col = flat % D;
        shortx8 kv_ = *reinterpret_cast<const shortx8*>(kg + kvbase + (long long)(kv0 + row) * k_ss + col);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          *reinterpret_cast<short*>(kt_lds + lds_off(col + i, row * 2, TR_RS)) = kv_[i];
        }
      }
    } else {
      constexpr int elems = KB * D;
      for (int flat = tid * 8; flat < elems; flat += NT * 8) {
        int row = flat / D, col = flat % D;
        shortx8 kv_, vv;
        if (kv0 + row < Skv) {
          kv_ = *reinterpret_cast<const shortx8*>(kg + kvbase + (long long)(kv0 + row) * k_ss + col);
          vv = *reinterpret_cast<const shortx8*>(vg + kvbase + (long long)(kv0 + row) * k_ss + col);
        } else {
          for (int i = 0; i < 8; ++i) { kv_[i] = 0; vv[i] = 0; }
        }
        *reinterpret_cast<shortx8*>(k_lds + lds_off(row, col * 2, NAT_RS)) = kv_;
        *reinterpret_cast<shortx8*>(v_lds + lds_off(row, col * 2, NAT_RS)) = vv;
        const int rot = threadIdx.x & 7;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          *reinterpret_cast<short*>(kt_lds + lds_off(col + i, row * 2, TR_RS)) = kv_[i];
        }
      }
    }
    __syncthreads();

    const int q_abs0 = q0 + wq * 16 + lg * 4;
    floatx4 ds[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      floatx4 s = {0.f, 0.f, 0.f, 0.f}, dp = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < NKK; ++kk) {
        shortx8 kfr = lds_read8(k_lds, nt * 16 + l16, kk * 32 + lg * 8, NAT_RS);
        s = mfma_bf16(qf[kk], kfr, s);
        shortx8 vfr = lds_read8(v_lds, nt * 16 + l16, kk * 32 + lg * 8, NAT_RS);
        dp = mfma_bf16(dof[kk], vfr, dp);
      }
      const int kv_abs = kv0 + nt * 16 + l16;
      const bool bnd = (kv0 + KB > Skv) || (CAUSAL && kv0 + KB > q_abs0);
      if (bnd) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float p = __expf(s[r] * scale - lse_r[r]);
          if (kv_abs >= Skv || (CAUSAL && kv_abs > q_abs0 + r)) p = 0.f;
          ds[nt][r] = p * (dp[r] - delta_r[r]) * scale;
        }
      } else {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float p = __expf(s[r] * scale - lse_r[r]);
          ds[nt][r] = p * (dp[r] - delta_r[r]) * scale;
        }
      }
    }

    char* pw = ps_lds + wq * (16 * KB * 2);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        *reinterpret_cast<short*>(pw + lds_off(lg * 4 + r, (nt * 16 + l16) * 2, KB * 2)) =
            f2bf(ds[nt][r]);
    LGKM0();

#pragma unroll
    for (int kk = 0; kk < KB / 32; ++kk) {
      shortx8 pa = lds_read8(pw, l16, kk * 32 + lg * 8, KB * 2);
#pragma unroll
      for (int dt = 0; dt < NDT; ++dt) {
        shortx8 bfr = lds_read8(kt_lds, dt * 16 + l16, kk * 32 + lg * 8, TR_RS);
        dq_acc[dt] = mfma_bf16(pa, bfr, dq_acc[dt]);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = q0 + wq * 16 + lg * 4 + r;
    if (row >= Sq) continue;
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt)
      dqg[dqbase + (long long)row * dq_ss + dt * 16 + l16] = f2bf(dq_acc[dt][r]);
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------
void flash_attn_fwd(const void* q, const void* k, const void* v, void* o,
                    float* lse, int64_t b, int64_t h, int64_t hkv, int64_t sq,
                    int64_t skv, int64_t dh, float scale, bool causal,
                    const int64_t* qs, const int64_t* ks, const int64_t* os,
                    hipStream_t s) {
  dim3 grid((unsigned)cdiv((int)sq, 128), (unsigned)(b * h));
  dim3 blk(512);
#define FAF(D, C)                                                             \
  hipLaunchKernelGGL((fa_fwd_kernel<D, C>), grid, blk, 0, s, (const short*)q, \
                     (const short*)k, (const short*)v, (short*)o, lse,        \
                     (int)b, (int)h, (int)hkv, (int)sq, (int)skv, scale,      \
                     qs[0], qs[1], qs[2], ks[0], ks[1], ks[2],                \
                     os[0], os[1], os[2])
  if (dh == 128) { if (causal) FAF(128, true); else FAF(128, false); }
  else           { if (causal) FAF(64, true);  else FAF(64, false); }
#undef FAF
}

void flash_attn_bwd(const void* dout, const void* q, const void* k,
                    const void* v, const void* o, const float* lse,
                    void* dq, void* dk, void* dv, float* delta, int64_t b,
                    int64_t h, int64_t hkv, int64_t sq, int64_t skv,
                    int64_t dh, float scale, bool causal,
                    const int64_t* qs, const int64_t* ks, const int64_t* dos,
                    const int64_t* os, const int64_t* dqs, const int64_t* dks,
                    const void* mask, const int64_t* ms, float pdrop,
                    uint64_t seed, uint64_t offset, hipStream_t s) {
  long long rows = b * h * sq;
  dim3 dgrid((unsigned)hmin<long long>(2048LL, (rows + 3) / 4));
  if (dh == 128)
    hipLaunchKernelGGL((fa_bwd_delta_kernel<128>), dgrid, dim3(256), 0, s,
                       (const short*)dout, (const short*)o, delta, (int)h,
                       (int)sq, rows, dos[0], dos[1], dos[2], os[0], os[1], os[2]);
  else
    hipLaunchKernelGGL((fa_bwd_delta_kernel<64>), dgrid, dim3(256), 0, s,
                       (const short*)dout, (const short*)o, delta, (int)h,
                       (int)sq, rows, dos[0], dos[1], dos[2], os[0], os[1], os[2]);

  dim3 gkv((unsigned)(b * h), (unsigned)cdiv((int)skv, 128));
  dim3 gq((unsigned)cdiv((int)sq, 128), (unsigned)(b * h));
  // The 16x16 dKV kernel is the default: both 32x32 rewrites measured
  // slower (combined/transposed-output: 146 TF bwd at occupancy 1;
  // occupancy-2 split with global A-frags: 95 TF; 16x16 baseline: 168 TF).
  // PA_FA_DKV32=1 opts into the 32x32 path for future experiments.
  static const bool dkv16_env = [] {
    const char* e = getenv("PA_FA_DKV32");
    return !(e && e[0] == '1');
  }();
  // the 16x16 dKV kernel has no mask/dropout path; route those to dkv32
  const bool use_dkv16 = dkv16_env && mask == nullptr && pdrop <= 0.f;
#define FAB(D, C)                                                              \
  do {                                                                         \
    if (use_dkv16)                                                             \
      hipLaunchKernelGGL((fa_bwd_dkv_kernel<D, C>), gkv, dim3(512), 0, s,      \
                         (const short*)dout, (const short*)q, (const short*)k, \
                         (const short*)v, lse, delta, (short*)dk, (short*)dv,  \
                         (int)b, (int)h, (int)sq, (int)skv, scale,             \
                         qs[0], qs[1], qs[2], ks[0], ks[1], ks[2],             \
                         dos[0], dos[1], dos[2], dks[0], dks[1], dks[2]);      \
    else                                                                       \
      flash_attn_bwd_dkv32(dout, q, k, v, lse, delta, dk, dv, b, h, sq, skv,   \
                           dh, scale, causal, qs, ks, dos, dks, mask, ms,      \
                           pdrop, seed, offset, s);                            \
    flash_attn_bwd_dq32(dout, q, k, v, lse, delta, dq, b, h, sq, skv, dh,     \
                        scale, causal, qs, ks, dos, dqs, mask, ms, pdrop,      \
                        seed, offset, s);                                      \
  } while (0)
  if (dh == 128) { if (causal) FAB(128, true); else FAB(128, false); }
  else           { if (causal) FAB(64, true);  else FAB(64, false); }
#undef FAB
}

void flash_attn_varlen_bwd(const void* dout, const void* q, const void* k,
                           const void* v, const void* o, const float* lse,
                           void* dq, void* dk, void* dv, float* delta,
                           int64_t h, int64_t total_q, int64_t total_k,
                           int64_t dh, float scale, bool causal,
                           int64_t nqblocks, int64_t nkvblocks,
                           const int* cu_q, const int* cu_k, const int* qbmap,
                           const int* kvbmap, float pdrop, uint64_t seed,
                           uint64_t offset, hipStream_t s) {
  // delta[h, total_q] = rowsum(do * o) over the packed [total, H, D] layout
  long long rows = h * total_q;
  dim3 dgrid((unsigned)hmin<long long>(2048LL, (rows + 3) / 4));
  const long long ss = h * dh;
  if (dh == 128)
    hipLaunchKernelGGL((fa_bwd_delta_kernel<128>), dgrid, dim3(256), 0, s,
                       (const short*)dout, (const short*)o, delta, (int)h,
                       (int)total_q, rows, 0, dh, ss, 0, dh, ss);
  else
    hipLaunchKernelGGL((fa_bwd_delta_kernel<64>), dgrid, dim3(256), 0, s,
                       (const short*)dout, (const short*)o, delta, (int)h,
                       (int)total_q, rows, 0, dh, ss, 0, dh, ss);
  flash_attn_varlen_bwd32(dout, q, k, v, lse, delta, dq, dk, dv, h, total_q,
                          total_k, dh, scale, causal, nqblocks, nkvblocks,
                          cu_q, cu_k, qbmap, kvbmap, pdrop, seed, offset, s);
}

}  // namespace pa
