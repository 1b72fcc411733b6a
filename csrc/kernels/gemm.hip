// Hand-written bf16 MFMA GEMM for gfx950 (C = op(A) x op(B), fp32
// accumulate, bf16 out) with fused epilogues.
//
// Reference role: paddle/phi/kernels/funcs/blas/blaslt_impl.cu.h +
// matmul_kernel_impl.h:108 (library GEMM), fused_gemm_epilogue_kernel.cu
// (bias+gelu epilogue), fused_linear_param_grad_add_kernel.cu (wgrad
// accumulate) -- all re-derived as native CDNA4 kernels.  A per-shape
// autotune table (paddle_amd/ops/gemm_table.json, the reference's
// matmul_kernel_impl.h:914-958 autotune-between-impls pattern) decides
// own-GEMM vs hipBLASLt per (layout, M, N, K).
//
// Geometry (guide "256^2 8-phase template", counted-vmcnt schedule):
//   * block tile 256x256, K-step 64, 8 waves (2M x 4N), 512 threads
//   * per-wave output 128x64 = acc[8][4] mfma_f32_16x16x32_bf16 frags
//   * LDS: double-buffered A[256][64] + B^T[256][64] bf16 = 128 KiB
//     (1 block/CU -- latency hidden by the phase interleave, not TLP)
//   * 4 phases per K-tile, one C-quadrant (16 MFMA) each, with
//     s_setprio(1) around the MFMA cluster (T5) and two barriers per
//     phase; all ds_reads for the tile issued in phases 0-1 and held in
//     VGPRs (A 64 + B 32 + acc 128 regs)
//   * staging for tile t+1 issued during tile t (A at phase 0, B at
//     phase 1) into the other LDS buffer; the ONLY vmcnt wait is a
//     counted s_waitcnt vmcnt(4) at phase 0 -- prefetch loads stay in
//     flight across barriers (T3+T4; never vmcnt(0) in the main loop)
//   * st_16x32-style XOR swizzle on LDS rows (T2); XCD-aware bijective
//     workgroup swizzle (T1)
//
// Operand layouts (all producing LDS A[m][k], B'[n][k]):
//   NT: A[M][K] rm, Bt[N][K] rm      (fwd X.W^T; both async-DMA staged)
//   NN: A[M][K] rm, B[K][N] rm       (dgrad dY.W; B reg-transposed)
//   TN: At[K][M] rm, B[K][N] rm      (wgrad dY^T.X; both reg-transposed)
// Epilogues: NONE / BIAS / BIAS_GELU(+aux out) / DGELU(aux in) / ACC.
#include "common.h"
#include "api.h"

namespace pa {

#define GEMM_SETPRIO(x) __builtin_amdgcn_s_setprio(x)

// LDS element offset with row-XOR swizzle: rows are 64 bf16 = 128 B; XOR
// the 16B-granule index (elem bits 3..5) with row&7 so a 16-lane column
// slice spreads over 8 bank-slots.
__device__ __forceinline__ unsigned g_swz(unsigned row, unsigned col) {
  return row * 64 + (col ^ ((row & 7u) << 3));
}

enum { LAY_NT = 0, LAY_NN = 1, LAY_TN = 2 };
enum { EPI_NONE = 0, EPI_BIAS = 1, EPI_BIAS_GELU = 2, EPI_DGELU = 3 };

__device__ __forceinline__ float epi_gelu(float x) {
  // exact-erf GeLU (paddle default approximate=False), same poly as
  // elementwise.hip so fused and unfused paths match bitwise-close
  float z = x * 0.70710678118654752440f;
  float az = fabsf(z);
  float t = 1.f / (1.f + 0.3275911f * az);
  float p = t * (0.254829592f + t * (-0.284496736f + t * (1.421413741f +
            t * (-1.453152027f + t * 1.061405429f))));
  float r = 1.f - p * __expf(-az * az);
  float e = copysignf(r, z);
  return 0.5f * x * (1.f + e);
}
__device__ __forceinline__ float epi_gelu_grad(float x) {
  float z = x * 0.70710678118654752440f;
  float az = fabsf(z);
  float t = 1.f / (1.f + 0.3275911f * az);
  float p = t * (0.254829592f + t * (-0.284496736f + t * (1.421413741f +
            t * (-1.453152027f + t * 1.061405429f))));
  float r = 1.f - p * __expf(-az * az);
  float e = copysignf(r, z);
  float cdf = 0.5f * (1.f + e);
  float pdf = 0.3989422804014327f * __expf(-0.5f * x * x);
  return cdf + x * pdf;
}

// ---------------------------------------------------------------------------
// the kernel
// ---------------------------------------------------------------------------
// FAST=true: interior-only instantiation (grid covers full 256x256 tiles,
// K%64==0; no guarded code compiled in -> no spills).  FAST=false: guarded
// boundary kernel launched over the full grid, early-exiting blocks the
// fast launch already covered (skip_interior).
template <int LAYOUT, int EPI, bool ACC, bool FAST>
__launch_bounds__(512, 2)
__global__ void gemm_bf16_kernel(const short* __restrict__ ag, const short* __restrict__ bg,
                                 short* __restrict__ cg, const short* __restrict__ biasg,
                                 short* __restrict__ auxg, int M, int N, int K,
                                 long long lda, long long ldb, long long ldc,
                                 int skip_interior) {
  constexpr int BM = 256, BN = 256, BK = 64;
  constexpr bool A_DMA = (LAYOUT != LAY_TN);   // A[M][K] row-major
  constexpr bool B_DMA = (LAYOUT == LAY_NT);   // Bt[N][K] row-major
  __shared__ short a_lds[2][BM * BK];
  __shared__ short b_lds[2][BN * BK];

  // T1: XCD-aware bijective workgroup swizzle (contiguous chunks per XCD)
  const int nwg = gridDim.x * gridDim.y;
  int orig = blockIdx.y * gridDim.x + blockIdx.x;
  {
    const int nx = 8;
    int q = nwg / nx, rr = nwg % nx;
    int xcd = orig % nx, pos = orig / nx;
    orig = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + pos;
  }
  // FAST grid spans only the interior tile rectangle
  const int mt = FAST ? (M / BM) : ((M + BM - 1) / BM);
  const int bm = orig % mt;
  const int bn = orig / mt;
  const int row0 = bm * BM, col0 = bn * BN;
  if (!FAST && skip_interior &&
      row0 + BM <= M && col0 + BN <= N && (K % BK) == 0)
    return;  // covered by the fast launch

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;          // 0..1: which 128 rows of C
  const int wc = wid & 3;           // 0..3: which 64 cols of C
  const int l16 = lane & 15;
  const int lg = lane >> 4;

  // DMA staging thread->element map: 4 rounds x (512 thr x 8 elem) per
  // 256x64 tile; round it covers rows [it*64, it*64+64)
  int s_r[4], s_cp[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    int flat = it * 4096 + tid * 8;
    int r = flat / BK, c = flat % BK;
    s_r[it] = r;
    s_cp[it] = c ^ ((r & 7) << 3);   // pre-swizzled source column
  }

  const int nkt = (K + BK - 1) / BK;

  // ---- staging lambdas ----------------------------------------------------
  // A operand, DMA path (A[M][K] row-major; NT/NN)
  auto stage_a_dma = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const short* src = ag + (long long)(row0 + s_r[it]) * lda + k0 + s_cp[it];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)&a_lds[buf][it * 4096 + tid * 8],
          16, 0, 0);
    }
  };
  // B operand, DMA path (Bt[N][K] row-major; NT)
  auto stage_b_dma = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const short* src = bg + (long long)(col0 + s_r[it]) * ldb + k0 + s_cp[it];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)&b_lds[buf][it * 4096 + tid * 8],
          16, 0, 0);
    }
  };
  // reg-transpose staging, load half: source S[K][X] row-major, 64 k-rows x
  // 256 x-cols slab -> regs (4 x shortx8); used for B (NN/TN) and A (TN)
  const int rot = tid & 7;
  auto load_trans = [&](const short* sg, long long lds_, int x0, int kt, shortx8* v) {
    const long long k0 = (long long)kt * BK;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int flat = it * 4096 + tid * 8;
      int kr = flat / BN;             // 0..63  (k within tile)
      int nc = flat % BN;             // 0..255 (x within tile)
      v[it] = *reinterpret_cast<const shortx8*>(sg + (k0 + kr) * lds_ + x0 + nc);
    }
  };
  // write the transposed slab into LDS[x][k] (rotated to spread banks)
  auto write_trans = [&](short* lds, const shortx8* v) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int flat = it * 4096 + tid * 8;
      int kr = flat / BN;
      int nc = flat % BN;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int i = (j + rot) & 7;
        lds[g_swz(nc + i, kr)] = v[it][i];
      }
    }
  };

  // ---- guarded (boundary / K-tail) staging: zero-fill out of range --------
  auto stage_guarded = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
    // A
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int flat = it * 4096 + tid * 8;
      int r = flat / BK, c = flat % BK;
      shortx8 v;
      if (A_DMA) {
        if (row0 + r < M && k0 + c + 7 < K)
          v = *reinterpret_cast<const shortx8*>(ag + (long long)(row0 + r) * lda + k0 + c);
        else
#pragma unroll
          for (int j = 0; j < 8; ++j)
            v[j] = (row0 + r < M && k0 + c + j < K)
                       ? ag[(long long)(row0 + r) * lda + k0 + c + j] : (short)0;
        *reinterpret_cast<shortx8*>(&a_lds[buf][g_swz(r, c)]) = v;
      }
    }
    if (!A_DMA) {
      // At[K][M]: read k-rows, scatter-transpose into a_lds[m][k]
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int flat = it * 4096 + tid * 8;
        int kr = flat / BN, nc = flat % BN;
        shortx8 v;
        if (k0 + kr < K && row0 + nc + 7 < M)
          v = *reinterpret_cast<const shortx8*>(ag + (k0 + kr) * lda + row0 + nc);
        else
#pragma unroll
          for (int j = 0; j < 8; ++j)
            v[j] = (k0 + kr < K && row0 + nc + j < M)
                       ? ag[(k0 + kr) * lda + row0 + nc + j] : (short)0;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          a_lds[buf][g_swz(nc + i, kr)] = v[i];
        }
      }
    }
    // B
    if (B_DMA) {
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int flat = it * 4096 + tid * 8;
        int r = flat / BK, c = flat % BK;
        shortx8 v;
        if (col0 + r < N && k0 + c + 7 < K)
          v = *reinterpret_cast<const shortx8*>(bg + (long long)(col0 + r) * ldb + k0 + c);
        else
#pragma unroll
          for (int j = 0; j < 8; ++j)
            v[j] = (col0 + r < N && k0 + c + j < K)
                       ? bg[(long long)(col0 + r) * ldb + k0 + c + j] : (short)0;
        *reinterpret_cast<shortx8*>(&b_lds[buf][g_swz(r, c)]) = v;
      }
    } else {
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int flat = it * 4096 + tid * 8;
        int kr = flat / BN, nc = flat % BN;
        shortx8 v;
        if (k0 + kr < K && col0 + nc + 7 < N)
          v = *reinterpret_cast<const shortx8*>(bg + (k0 + kr) * ldb + col0 + nc);
        else
#pragma unroll
          for (int j = 0; j < 8; ++j)
            v[j] = (k0 + kr < K && col0 + nc + j < N)
                       ? bg[(k0 + kr) * ldb + col0 + nc + j] : (short)0;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          b_lds[buf][g_swz(nc + i, kr)] = v[i];
        }
      }
    }
  };

  floatx4 acc[8][4];
#pragma unroll
  for (int m = 0; m < 8; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  // fragment registers, reused across phases (guide's 12/4/8/4 read split):
  //   af2 holds the current m-half of A (m0 for phases 0-1, m1 for 2-3)
  //   bf2 holds the current n-half of B (n0, n1, n1, n0 re-read)
  shortx8 af2[4][2];  // [mfrag within half][ks]
  shortx8 bf2[2][2];  // [nfrag within half][ks]

  auto read_a_half = [&](int buf, int mh) {
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        af2[m][ks] = *reinterpret_cast<const shortx8*>(
            &a_lds[buf][g_swz(wr * 128 + (mh * 4 + m) * 16 + l16, ks * 32 + lg * 8)]);
  };
  auto read_b_half = [&](int buf, int nh) {
#pragma unroll
    for (int n = 0; n < 2; ++n)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bf2[n][ks] = *reinterpret_cast<const shortx8*>(
            &b_lds[buf][g_swz(wc * 64 + (nh * 2 + n) * 16 + l16, ks * 32 + lg * 8)]);
  };
  auto mfma_quadrant = [&](int mh, int nh) {
    GEMM_SETPRIO(1);
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 2; ++n)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[mh * 4 + m][nh * 2 + n] =
              mfma_bf16(af2[m][ks], bf2[n][ks], acc[mh * 4 + m][nh * 2 + n]);
    GEMM_SETPRIO(0);
  };

  if constexpr (FAST) {
    // =======================================================================
    // fast path: 4-phase counted-vmcnt pipeline (T3+T4), T5 setprio
    // =======================================================================
    shortx8 breg[4];  // reg-staged B slab (NN/TN)
    shortx8 areg[4];  // reg-staged A slab (TN)

    // prologue: stage tile 0 into buf 0
    if (!A_DMA) load_trans(ag, lda, row0, 0, areg);
    if (!B_DMA) load_trans(bg, ldb, col0, 0, breg);
    if (A_DMA) stage_a_dma(0, 0);
    if (B_DMA) stage_b_dma(0, 0);
    if (!B_DMA || !A_DMA) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      if (!A_DMA) write_trans(&a_lds[0][0], areg);
      if (!B_DMA) write_trans(&b_lds[0][0], breg);
    }
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __syncthreads();

    for (int kt = 0; kt < nkt; ++kt) {
      const int cur = kt & 1;
      const bool pf = (kt + 1 < nkt);

      // ---- phase 0: quadrant (m0,n0); issue next-tile loads ----
      read_a_half(cur, 0);
      read_b_half(cur, 0);
      if (pf) {
        if (!A_DMA) load_trans(ag, lda, row0, kt + 1, areg);         // TN A
        if (!B_DMA && A_DMA) load_trans(bg, ldb, col0, kt + 1, breg);  // NN B
        if (A_DMA) stage_a_dma(cur ^ 1, kt + 1);
      }
      // counted wait for THIS tile's data: prefetch loads stay in flight.
      // On the last tile nothing newer was issued, so drain fully.
      if (A_DMA && B_DMA) {
        if (pf) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      } else if (A_DMA) {
        if (pf) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
        else    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      // TN: LDS was filled by ds_writes, drained at phase 2's lgkmcnt(0)
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(0, 0);
      __builtin_amdgcn_s_barrier();

      // ---- phase 1: quadrant (m0,n1); write-transpose A (TN) / B (NN) ----
      // Writing buf^1 here is safe: tile kt-1's last reads of it (its n0
      // re-read) were lgkm-drained before kt-1's closing barrier.
      read_b_half(cur, 1);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (pf && B_DMA) stage_b_dma(cur ^ 1, kt + 1);
      if (pf && !A_DMA) {
        if (!B_DMA) load_trans(bg, ldb, col0, kt + 1, breg);  // TN B issue
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");      // areg landed
        write_trans(&a_lds[cur ^ 1][0], areg);
      } else if (pf && !B_DMA) {
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");      // breg landed
        write_trans(&b_lds[cur ^ 1][0], breg);                // NN
      }
      __builtin_amdgcn_s_barrier();
      mfma_quadrant(0, 1);
      __builtin_amdgcn_s_barrier();

      // ---- phase 2: quadrant (m1,n1); write-transpose B (TN) ----
      read_a_half(cur, 1);
      if (pf && !B_DMA && !A_DMA) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        write_trans(&b_lds[cur ^ 1][0], breg);
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // A(m1) + our writes
      mfma_quadrant(1, 1);
      __builtin_amdgcn_s_barrier();

      // ---- phase 3: quadrant (m1,n0); re-read B(n0) ----
      read_b_half(cur, 0);
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(1, 0);
      __builtin_amdgcn_s_barrier();
    }
  } else {
    // =======================================================================
    // boundary path: guarded staging, conservative waits
    // =======================================================================
    stage_guarded(0, 0);
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __syncthreads();
    for (int kt = 0; kt < nkt; ++kt) {
      const int cur = kt & 1;
      if (kt + 1 < nkt) stage_guarded(cur ^ 1, kt + 1);
      read_a_half(cur, 0);
      read_b_half(cur, 0);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(0, 0);
      read_b_half(cur, 1);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(0, 1);
      read_a_half(cur, 1);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(1, 1);
      read_b_half(cur, 0);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(1, 0);
      asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
      __syncthreads();
    }
  }

  // ---- epilogue -----------------------------------------------------------
  // bias per output column (bf16), loaded once per nfrag
  float bias_v[4];
  if (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int col = col0 + wc * 64 + nf * 16 + l16;
      bias_v[nf] = (col < N && biasg) ? bf2f(biasg[col]) : 0.f;
    }
  }
#pragma unroll
  for (int m = 0; m < 8; ++m) {
    int row = row0 + wr * 128 + m * 16 + (lg * 4);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (row + r >= M) continue;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int col = col0 + wc * 64 + nf * 16 + l16;
        if (col >= N) continue;
        long long idx = (long long)(row + r) * ldc + col;
        float v = acc[m][nf][r];
        if (EPI == EPI_BIAS) {
          v += bias_v[nf];
        } else if (EPI == EPI_BIAS_GELU) {
          v += bias_v[nf];
          if (auxg) auxg[idx] = f2bf(v);
          v = epi_gelu(v);
        } else if (EPI == EPI_DGELU) {
          v *= epi_gelu_grad(bf2f(auxg[idx]));
        }
        if (ACC) v += bf2f(cg[idx]);
        cg[idx] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host launcher
// ---------------------------------------------------------------------------
void gemm_bf16_ex(const void* a, const void* b, void* c, const void* bias,
                  void* aux, int64_t m, int64_t n, int64_t k, int64_t lda,
                  int64_t ldb, int64_t ldc, int layout, int epilogue,
                  bool accumulate, hipStream_t s) {
  const int mi = (int)(m / 256), ni = (int)(n / 256);      // interior tiles
  const int gm = (int)((m + 255) / 256), gn = (int)((n + 255) / 256);
  const bool k_ok = (k % 64 == 0);
  const bool has_fast = k_ok && mi > 0 && ni > 0;
  const bool has_edge = !k_ok || mi < gm || ni < gn;
  dim3 blk(512);
#define LAUNCH(L, E, AC)                                                       \
  do {                                                                         \
    if (has_fast)                                                              \
      hipLaunchKernelGGL((gemm_bf16_kernel<L, E, AC, true>),                   \
                         dim3((unsigned)mi, (unsigned)ni), blk, 0, s,          \
                         (const short*)a, (const short*)b, (short*)c,          \
                         (const short*)bias, (short*)aux, (int)m, (int)n,      \
                         (int)k, lda, ldb, ldc, 0);                            \
    if (has_edge)                                                              \
      hipLaunchKernelGGL((gemm_bf16_kernel<L, E, AC, false>),                  \
                         dim3((unsigned)gm, (unsigned)gn), blk, 0, s,          \
                         (const short*)a, (const short*)b, (short*)c,          \
                         (const short*)bias, (short*)aux, (int)m, (int)n,      \
                         (int)k, lda, ldb, ldc, has_fast ? 1 : 0);             \
  } while (0)
#define EPI_SWITCH(L)                                                        \
  do {                                                                       \
    if (accumulate) { LAUNCH(L, EPI_NONE, true); }                           \
    else if (epilogue == EPI_BIAS) { LAUNCH(L, EPI_BIAS, false); }           \
    else if (epilogue == EPI_BIAS_GELU) { LAUNCH(L, EPI_BIAS_GELU, false); } \
    else if (epilogue == EPI_DGELU) { LAUNCH(L, EPI_DGELU, false); }         \
    else { LAUNCH(L, EPI_NONE, false); }                                     \
  } while (0)
  if (layout == LAY_NT) EPI_SWITCH(LAY_NT);
  else if (layout == LAY_NN) EPI_SWITCH(LAY_NN);
  else EPI_SWITCH(LAY_TN);
#undef EPI_SWITCH
#undef LAUNCH
}

// legacy entry (round-1 API): b_is_nt picks NT vs NN, no epilogue
void gemm_bf16(const void* a, const void* b, void* c, int64_t m, int64_t n,
               int64_t k, int64_t lda, int64_t ldb, int64_t ldc, bool b_is_nt,
               hipStream_t s) {
  gemm_bf16_ex(a, b, c, nullptr, nullptr, m, n, k, lda, ldb, ldc,
               b_is_nt ? LAY_NT : LAY_NN, EPI_NONE, false, s);
}

}  // namespace pa
