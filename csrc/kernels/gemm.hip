// Hand-written bf16 MFMA GEMM for gfx950 (C = A x B, fp32 accumulate,
// bf16 out).
//
// Geometry: block tile 256x256, K-step 64 (two 32-wide k-half phases),
// 8 waves (2M x 4N, per-wave C = 128x64 = acc[8][4] of
// mfma_f32_16x16x32_bf16).
//
// Pipeline (the load-bearing part, guide §5.5 T3/T4): LDS is a RING of
// 10 half-tile slots (16 KiB each = [256 rows][32 k] bf16) giving
// 2-K-tile lookahead.  Each phase stages one K-tile-half pair for tile
// kt+2 via async global_load_lds (pre-swizzled source), computes 32
// MFMAs, then waits a COUNTED `s_waitcnt vmcnt(16)` -- staged loads stay
// in flight across barriers instead of draining (the m97-ceiling
// mistake), followed by a raw s_barrier.  s_setprio(1) wraps the MFMA
// cluster (T5); XCD-aware bijective workgroup swizzle (T1/m204).
//
// Layouts: NT: A[M][K] rm + Bt[N][K] rm (both DMA-staged, fast).
//          NN: A[M][K] rm + B[K][N] rm (B transposed via reg staging).
#include "common.h"
#include "api.h"

namespace pa {

#define RAW_BARRIER() __builtin_amdgcn_s_barrier()
#define WAITCNT_VM(N) asm volatile("s_waitcnt vmcnt(" #N ")" ::: "memory")
#define WAITCNT_LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")

// half-tile slot: [256][32] bf16; row stride 64B.  XOR the 8-elem granule
// (bits 3..4) with row&3: a 16-lane column-slice read spreads over
// (row parity x 4 granules) = 8 bank-slots -> ~2-way.
__device__ __forceinline__ unsigned h_swz(unsigned row, unsigned col) {
  return row * 32 + (col ^ ((row & 3u) << 3));
}

template <bool BT>
__launch_bounds__(512, 2)
__global__ void gemm_bf16_kernel(const short* __restrict__ ag, const short* __restrict__ bg,
                                 short* __restrict__ cg, int M, int N, int K,
                                 long long lda, long long ldb, long long ldc) {
  constexpr int BM = 256, BN = 256, BK = 64;
  constexpr int NSLOT = 10;
  __shared__ short ring[NSLOT][256 * 32];

  const int nwg = gridDim.x * gridDim.y;
  int orig = blockIdx.y * gridDim.x + blockIdx.x;
  {
    const int nx = 8;
    int q = nwg / nx, rr = nwg % nx;
    int xcd = orig % nx, pos = orig / nx;
    orig = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + pos;
  }
  const int gm = (M + BM - 1) / BM;
  const int bm = orig % gm;
  const int bn = orig / gm;
  const int row0 = bm * BM, col0 = bn * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;
  const int wc = wid & 3;
  const int l16 = lane & 15;
  const int lg = lane >> 4;

  // staging map: half-tile 256x32 = 8192 elems; 512thr x 8 = 2 rounds
  int s_r[2], s_cp[2];
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    int flat = it * 4096 + tid * 8;
    int r = flat / 32, c = flat % 32;
    s_r[it] = r;
    s_cp[it] = c ^ ((r & 3) << 3);
  }

  const bool interior = (row0 + BM <= M) && (col0 + BN <= N) && (K % BK == 0);

  // slot index for half h of tile kt: h_global = kt*4 + idx,
  // idx: 0=A_k0 1=B_k0 2=A_k1 3=B_k1
  auto slot = [&](int kt, int idx) { return (kt * 4 + idx) % NSLOT; };

  // stage one A half (k-half kh) of tile kt into its slot (2 DMA / thread)
  auto stage_a = [&](int kt, int kh) {
    const long long k0 = (long long)kt * BK + kh * 32;
    short* dst = ring[slot(kt, kh * 2)];
    if (interior) {
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        const short* src = ag + (long long)(row0 + s_r[it]) * lda + k0 + s_cp[it];
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)src,
            (__attribute__((address_space(3))) unsigned int*)&dst[it * 4096 + tid * 8],
            16, 0, 0);
      }
    } else {
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        int r = s_r[it], c = (it * 4096 + tid * 8) % 32;
        shortx8 v;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = (row0 + r < M && k0 + c + j < K)
                     ? ag[(long long)(row0 + r) * lda + k0 + c + j] : (short)0;
        *reinterpret_cast<shortx8*>(&dst[h_swz(r, c)]) = v;
      }
    }
  };

  auto stage_b = [&](int kt, int kh) {
    const long long k0 = (long long)kt * BK + kh * 32;
    short* dst = ring[slot(kt, kh * 2 + 1)];
    if (BT) {
      if (interior) {
#pragma unroll
        for (int it = 0; it < 2; ++it) {
          const short* src = bg + (long long)(col0 + s_r[it]) * ldb + k0 + s_cp[it];
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) unsigned int*)src,
              (__attribute__((address_space(3))) unsigned int*)&dst[it * 4096 + tid * 8],
              16, 0, 0);
        }
      } else {
#pragma unroll
        for (int it = 0; it < 2; ++it) {
          int r = s_r[it], c = (it * 4096 + tid * 8) % 32;
          shortx8 v;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            v[j] = (col0 + r < N && k0 + c + j < K)
                       ? bg[(long long)(col0 + r) * ldb + k0 + c + j] : (short)0;
          *reinterpret_cast<shortx8*>(&dst[h_swz(r, c)]) = v;
        }
      }
    } else {
      // B [K][N]: rows are 32 k x 256 n; transpose into [n][k] slot
      const int rot = tid & 7;
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        int flat = it * 4096 + tid * 8;
        int kr = flat / BN, nc = flat % BN;
        shortx8 v;
        if (interior || (k0 + kr < K && col0 + nc + 7 < N))
          v = *reinterpret_cast<const shortx8*>(bg + (k0 + kr) * ldb + col0 + nc);
        else
#pragma unroll
          for (int j = 0; j < 8; ++j)
            v[j] = (k0 + kr < K && col0 + nc + j < N)
                       ? bg[(k0 + kr) * ldb + col0 + nc + j] : (short)0;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int i = (j + rot) & 7;
          dst[h_swz(nc + i, kr)] = v[i];
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
  // DMA-count discipline only valid when every stage in this block is DMA
  const bool counted = interior && BT;  // NN's B staging is reg-based

  // prologue: tiles 0 and 1 fully staged
  stage_a(0, 0); stage_b(0, 0); stage_a(0, 1); stage_b(0, 1);
  if (nkt > 1) { stage_a(1, 0); stage_b(1, 0); stage_a(1, 1); stage_b(1, 1); }
  if (counted && nkt > 1) {
    WAITCNT_VM(12);  // tile0 phase-0 halves done; 3 newer stage-events fly
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  WAITCNT_LGKM0();
  RAW_BARRIER();

  for (int kt = 0; kt < nkt; ++kt) {
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      // stage tile kt+2's matching halves (slots freed by the barrier
      // that ended the previous phase -- ring arithmetic in header note)
      if (kt + 2 < nkt) {
        stage_a(kt + 2, kh);
        stage_b(kt + 2, kh);
      }
      const short* a_sl = ring[slot(kt, kh * 2)];
      const short* b_sl = ring[slot(kt, kh * 2 + 1)];
      shortx8 af[8], bf[4];
#pragma unroll
      for (int m = 0; m < 8; ++m)
        af[m] = *reinterpret_cast<const shortx8*>(
            &a_sl[h_swz(wr * 128 + m * 16 + l16, lg * 8)]);
#pragma unroll
      for (int n = 0; n < 4; ++n)
        bf[n] = *reinterpret_cast<const shortx8*>(
            &b_sl[h_swz(wc * 64 + n * 16 + l16, lg * 8)]);
      WAITCNT_LGKM0();
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m = 0; m < 8; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = mfma_bf16(af[m], bf[n], acc[m][n]);
      __builtin_amdgcn_s_setprio(0);
      // counted wait: the NEXT phase's halves were staged 2 tiles ago;
      // 3 stage-events (4 DMA each) were issued after them -> allow 12
      if (counted) {
        WAITCNT_VM(12);
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        WAITCNT_LGKM0();
      }
      RAW_BARRIER();
    }
  }

  // epilogue
#pragma unroll
  for (int m = 0; m < 8; ++m) {
    int row = row0 + wr * 128 + m * 16 + lg * 4;
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
