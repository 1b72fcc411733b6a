// Experimental fine-interleaved NT GEMM (guide's 256x256 8-phase class):
// 3-deep K-tile ring (BK=32) with COUNTED vmcnt at tile boundaries only,
// per-phase [ds_read | stage | barrier | lgkm(0) | setprio'd MFMA | barrier]
// interleave, 80 B padded LDS rows (bank-spread without swizzle).
//
// v1 restrictions: NT layout, M%256==0, N%256==0, K%32==0.
//
// MEASURED OUTCOME (kept as a documented negative result, not
// dispatched): 869/1076/1079 TF at 4k/8k/fc1 -- below the 4-phase
// kernel (1020/1236/1177) and hipBLASLt (1.3-1.6 PF).  Journey:
// VGPR-staged variant 732 TF (per-phase lgkmcnt(0) swallowed the global
// latency -- staging stores are lgkm-tracked); DMA staging with per-lane
// LDS dests corrupted results (global_load_lds dest is WAVE-UNIFORM +
// lane*16B); this version is correct with wave-span dests + pre-swizzled
// source granules.  Remaining gap vs the guide's 1563 TF template:
// BK=32 keeps global bursts at 64 B/row (the template's BK=64 half-tile
// staging reads 128 B rows) and the half-tile prefetch skew that makes
// the tile-boundary wait fully counted at BK=64 needs a consumption
// order this derivation could not reconstruct -- see STATUS.md round-3
// queue for the staging-ring design notes.
#include "common.h"
#include "api.h"

namespace pa {

__global__ __launch_bounds__(512, 1)
void gemm8_nt_kernel(const short* __restrict__ ag, const short* __restrict__ bg,
                     short* __restrict__ cg, int M, int N, int K,
                     long long lda, long long ldb, long long ldc) {
  constexpr int BM = 256, BN = 256, BK = 32;
  constexpr int ROWB = 64;      // linear 64 B rows; bank spread via the
                                // granule XOR swizzle (DMA's LDS dest is
                                // wave-uniform + lane*16B, so padded
                                // per-lane layouts can't be DMA targets)
  __shared__ char a_lds[3][BM * ROWB];
  __shared__ char b_lds[3][BN * ROWB];

  // XCD-bijective swizzle + GM rasterization (as gemm.hip)
  const int nwg = gridDim.x * gridDim.y;
  int orig = blockIdx.y * gridDim.x + blockIdx.x;
  {
    const int nx = 8;
    int q = nwg / nx, rr = nwg % nx;
    int xcd = orig % nx, pos = orig / nx;
    orig = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + pos;
  }
  const int mt = M / BM, nt = N / BN;
  constexpr int GM = 8;
  int bm, bn;
  {
    int band = orig / (GM * nt);
    int rem = orig - band * (GM * nt);
    int gm_band = mt - band * GM < GM ? mt - band * GM : GM;
    bm = band * GM + rem % gm_band;
    bn = rem / gm_band;
  }
  const int row0 = bm * BM, col0 = bn * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;        // 0..1 -> C rows [wm*128, +128)
  const int wn = wid & 3;         // 0..3 -> C cols [wn*64, +64)
  const int l16 = lane & 15;
  const int lg = lane >> 4;       // k-group (8 elems)

  floatx4 acc[8][4];
#pragma unroll
  for (int s = 0; s < 8; ++s)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[s][n] = floatx4{0.f, 0.f, 0.f, 0.f};

  // staging map: per DMA instruction a WAVE writes 1 KiB = 16 rows; the
  // lane's GLOBAL column is pre-swizzled so data lands at the granule-
  // XOR'd LDS position the fragment reads expect
  const int s_rowl = lane >> 2;         // row within the wave's 16-row span
  const int s_g = lane & 3;             // 16 B granule within the 64 B row

  // async DMA staging: global_load_lds keeps the transfer out of BOTH
  // the VGPR file and the lgkm counter, so the per-phase lgkmcnt(0)
  // waits only the fragment ds_reads (VGPR staging serialized the
  // global latency into every phase: 732 vs 1047 TF)
  auto stage = [&](int t) {             // stage K-tile t (both operands)
    const int slot = t % 3;
    const int kb = t * BK;              // element k offset
#pragma unroll
    for (int r = 0; r < 2; ++r) {
      int row16 = r * 8 + wid;          // which 16-row span
      int row = row16 * 16 + s_rowl;
      int gp = s_g ^ (row & 3);         // pre-swizzled source granule
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              ag + (long long)(row0 + row) * lda + kb + gp * 8),
          (__attribute__((address_space(3))) unsigned int*)(
              &a_lds[slot][row16 * 1024]), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              bg + (long long)(col0 + row) * ldb + kb + gp * 8),
          (__attribute__((address_space(3))) unsigned int*)(
              &b_lds[slot][row16 * 1024]), 16, 0, 0);
    }
  };

  const int nkt = K / BK;
  stage(0);
  if (nkt > 1) stage(1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < nkt; ++t) {
    const int slot = t % 3;
    // B-frags for the whole tile (4 n-subtiles)
    shortx8 bf[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int row = wn * 64 + n * 16 + l16;
      bf[n] = *reinterpret_cast<const shortx8*>(
          &b_lds[slot][row * ROWB + (lg ^ (row & 3)) * 16]);
    }
#pragma unroll
    for (int ph = 0; ph < 2; ++ph) {
      shortx8 af[4];
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        int row = wm * 128 + (ph * 4 + s) * 16 + l16;
        af[s] = *reinterpret_cast<const shortx8*>(
            &a_lds[slot][row * ROWB + (lg ^ (row & 3)) * 16]);
      }
      // stage half of tile t+2 (2 x 16 B per thread per phase)
      if (ph == 0 && t + 2 < nkt) stage(t + 2);
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[ph * 4 + s][n] = mfma_bf16(af[s], bf[n], acc[ph * 4 + s][n]);
      __builtin_amdgcn_s_setprio(0);
      // tile boundary: tile t+1 (staged during t-1) must have landed;
      // outstanding = this tile's 4 DMAs (for t+2) -> counted wait
      if (ph == 1) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue: C[row][col], lane holds rows lg*4+r of each 16x16 subtile
#pragma unroll
  for (int s = 0; s < 8; ++s)
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int col = col0 + wn * 64 + n * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wm * 128 + s * 16 + lg * 4 + r;
        cg[(long long)row * ldc + col] = f2bf(acc[s][n][r]);
      }
    }
}

void gemm_bf16_8p(const void* a, const void* b, void* c, int64_t m, int64_t n,
                  int64_t k, int64_t lda, int64_t ldb, int64_t ldc,
                  hipStream_t s) {
  dim3 grid((unsigned)(m / 256), (unsigned)(n / 256));
  hipLaunchKernelGGL(gemm8_nt_kernel, grid, dim3(512), 0, s, (const short*)a,
                     (const short*)b, (short*)c, (int)m, (int)n, (int)k,
                     lda, ldb, ldc);
}

}  // namespace pa
