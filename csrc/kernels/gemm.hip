// Hand-written bf16 MFMA GEMM for gfx950 (C = A x B, fp32 accumulate,
// bf16 out).
//
// Reference role: paddle/phi/kernels/funcs/blas/blaslt_impl.cu.h +
// matmul_kernel_impl.h:108 (the library-GEMM path) -- here re-derived as
// a native CDNA4 kernel; hipBLASLt remains the dispatch where it wins.
//
// Geometry (guide §5 "256² 8-phase template" -- re-derived, simplified
// schedule):
//   * block tile 256x256, K-step 64, 8 waves (2M x 4N), 512 threads
//   * per-wave output 128x64 = acc[8][4] mfma_f32_16x16x32_bf16 fragments
//   * LDS: double-buffered A[256][64] + B^T[256][64] bf16 = 128 KiB
//   * staging for K-tile t+1 issued at the START of tile t's compute via
//     async global_load_lds (pre-swizzled source) -- ~64 MFMA of cover
//     for the HBM latency (T14 issue-early at K-tile granularity)
//   * st_16x32-style XOR swizzle on LDS rows; s_setprio around the MFMA
//     cluster (T5); XCD-aware bijective workgroup swizzle (T1, m204)
//
// Operand layouts (templates):
//   NT: A[M][K] rm, Bt[N][K] rm  (fastest: both DMA-staged)
//   NN: A[M][K] rm, B[K][N] rm   (B transposed into LDS via reg staging)
// TN (wgrad) stays on hipBLASLt for now (python autotune picks per shape).
#include "common.h"
#include "api.h"

namespace pa {

typedef __attribute__((ext_vector_type(2))) int intx2;

#define GEMM_SETPRIO(x) __builtin_amdgcn_s_setprio(x)

// LDS element offset with row-XOR swizzle: rows are 64 bf16 = 128 B; XOR
// the 16B-granule index (elem bits 3..5) with row&7 so a 16-lane column
// slice spreads over 8 bank-slots (2-way; same fix as flash_attn.hip).
__device__ __forceinline__ unsigned g_swz(unsigned row, unsigned col) {
  return row * 64 + (col ^ ((row & 7u) << 3));
}

template <bool BT>  // BT=true: B supplied as Bt[N][K] (NT); false: B[K][N]
__launch_bounds__(512, 2)
__global__ void gemm_bf16_kernel(const short* __restrict__ ag, const short* __restrict__ bg,
                                 short* __restrict__ cg, int M, int N, int K,
                                 long long lda, long long ldb, long long ldc) {
  constexpr int BM = 256, BN = 256, BK = 64;
  __shared__ short a_lds[2][BM * BK];
  __shared__ short b_lds[2][BN * BK];

  // XCD-aware bijective workgroup swizzle (m204): contiguous chunks per XCD
  const int nwg = gridDim.x * gridDim.y;
  int orig = blockIdx.y * gridDim.x + blockIdx.x;
  {
    const int nx = 8;
    int q = nwg / nx, rr = nwg % nx;
    int xcd = orig % nx, pos = orig / nx;
    orig = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + pos;
  }
  const int bm = orig % ((M + BM - 1) / BM);
  const int bn = orig / ((M + BM - 1) / BM);
  const int row0 = bm * BM, col0 = bn * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;          // 0..1: which 128 rows
  const int wc = wid & 3;           // 0..3: which 64 cols
  const int l16 = lane & 15;
  const int lg = lane >> 4;

  // staging thread->element maps (once)
  // A tile: 256x64 = 16384 elems; 512 thr x 8 = 4096/round -> 4 rounds
  int a_r[4], a_cp[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    int flat = it * 4096 + tid * 8;
    int r = flat / BK, c = flat % BK;
    a_r[it] = r;
    a_cp[it] = c ^ ((r & 7) << 3);   // pre-swizzled source column
  }

  const bool interior = (row0 + BM <= M) && (col0 + BN <= N);

  auto stage = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
    if (interior && K - k0 >= BK) {
      // async DMA path
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        const short* src = ag + (long long)(row0 + a_r[it]) * lda + k0 + a_cp[it];
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)src,
            (__attribute__((address_space(3))) unsigned int*)&a_lds[buf][it * 4096 + tid * 8],
            16, 0, 0);
      }
      if (BT) {
#pragma unroll
        for (int it = 0; it < 4; ++it) {
          const short* src = bg + (long long)(col0 + a_r[it]) * ldb + k0 + a_cp[it];
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) unsigned int*)src,
              (__attribute__((address_space(3))) unsigned int*)&b_lds[buf][it * 4096 + tid * 8],
              16, 0, 0);
        }
      } else {
        // B [K][N]: load rows of B (contiguous n), write transposed to LDS
        const int rot = tid & 7;
#pragma unroll
        for (int it = 0; it < 4; ++it) {
          int flat = it * 4096 + tid * 8;
          int kr = flat / BN;             // 0..63  (k within tile)
          int nc = flat % BN;             // 0..255 (n within tile)
          shortx8 v = *reinterpret_cast<const shortx8*>(
              bg + (k0 + kr) * ldb + col0 + nc);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            int i = (j + rot) & 7;
            b_lds[buf][g_swz(nc + i, kr)] = v[i];
          }
        }
      }
    } else {
      // boundary tile: guarded reg staging (zero-fill)
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int flat = it * 4096 + tid * 8;
        int r = flat / BK, c = flat % BK;
        shortx8 v;
        if (row0 + r < M && k0 + c + 7 < K)
          v = *reinterpret_cast<const shortx8*>(ag + (long long)(row0 + r) * lda + k0 + c);
        else
#pragma unroll
          for (int j = 0; j < 8; ++j)
            v[j] = (row0 + r < M && k0 + c + j < K)
                       ? ag[(long long)(row0 + r) * lda + k0 + c + j] : (short)0;
        *reinterpret_cast<shortx8*>(&a_lds[buf][g_swz(r, c)]) = v;  // c%8==0 -> 16B aligned
      }
      if (BT) {
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
        const int rot = tid & 7;
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
    }
  };

  floatx4 acc[8][4];
#pragma unroll
  for (int m = 0; m < 8; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  const int nkt = (K + BK - 1) / BK;
  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int kt = 0; kt < nkt; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < nkt) stage(cur ^ 1, kt + 1);  // issue early: covered by MFMAs

    // ---- compute K-tile kt: 4 sub-phases of 2 mf x 4 nf x 2 ks ----------
    // B frags (shared across mf): load once
    shortx8 bf[4][2];
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bf[nf][ks] = *reinterpret_cast<const shortx8*>(
            &b_lds[cur][g_swz(wc * 64 + nf * 16 + l16, ks * 32 + lg * 8)]);
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      shortx8 af[2][2];
#pragma unroll
      for (int m = 0; m < 2; ++m)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          af[m][ks] = *reinterpret_cast<const shortx8*>(
              &a_lds[cur][g_swz(wr * 128 + (q * 2 + m) * 16 + l16, ks * 32 + lg * 8)]);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      GEMM_SETPRIO(1);
#pragma unroll
      for (int m = 0; m < 2; ++m)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[q * 2 + m][nf] = mfma_bf16(af[m][ks], bf[nf][ks], acc[q * 2 + m][nf]);
      GEMM_SETPRIO(0);
    }
    // next tile's loads must have landed before we flip buffers
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  // ---- epilogue: bf16 store -------------------------------------------
#pragma unroll
  for (int m = 0; m < 8; ++m) {
    int row = row0 + wr * 128 + m * 16 + (lg * 4);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (row + r >= M) continue;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int col = col0 + wc * 64 + nf * 16 + l16;
        if (col < N)
          cg[(long long)(row + r) * ldc + col] = f2bf(acc[m][nf][r]);
      }
    }
  }
}

void gemm_bf16(const void* a, const void* b, void* c, int64_t m, int64_t n,
               int64_t k, int64_t lda, int64_t ldb, int64_t ldc, bool b_is_nt,
               hipStream_t s) {
  int gm = (int)((m + 255) / 256), gn = (int)((n + 255) / 256);
  dim3 grid((unsigned)gm, (unsigned)gn);
  dim3 blk(512);
  if (b_is_nt)
    hipLaunchKernelGGL((gemm_bf16_kernel<true>), grid, blk, 0, s,
                       (const short*)a, (const short*)b, (short*)c, (int)m,
                       (int)n, (int)k, lda, ldb, ldc);
  else
    hipLaunchKernelGGL((gemm_bf16_kernel<false>), grid, blk, 0, s,
                       (const short*)a, (const short*)b, (short*)c, (int)m,
                       (int)n, (int)k, lda, ldb, ldc);
}

}  // namespace pa
