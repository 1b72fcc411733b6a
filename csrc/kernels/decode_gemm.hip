// Skinny decode GEMM for gfx950 serving: y[M,N] = x[M,K] @ W[K,N] + bias
// with M <= 32 (a decode micro-batch) -- the shape class where hipBLASLt's
// Tensile tiles reach only ~25% of HBM bandwidth (profiles/
// serve_kernel_stats.csv: 79.6 us for a 134 MB weight read).
//
// Reference role: the decoder GEMM path of fused_multi_transformer
// (paddle/phi/kernels/fusion/gpu/fused_multi_transformer_kernel.cu).
//
// Design: the weight stream IS the kernel -- W[K,N] row-major is read
// exactly once, 8 B/lane coalesced; x (<= 32x K) is staged per K-chunk in
// LDS and broadcast.  Split-K across blocks writes fp32 partials;
// a tiny second kernel reduces + adds bias + casts to bf16.
#include "common.h"
#include "api.h"

namespace pa {

// block: 256 threads, 4 cols/thread -> 1024 N-cols per block
// grid.x = ceil(N/1024), grid.y = ksplit; partial [ksplit, MT, N] fp32
template <int MT>
__launch_bounds__(256)
__global__ void decode_gemm_kernel(const short* __restrict__ xg,
                                   const short* __restrict__ wg,
                                   float* __restrict__ partial,
                                   int M, int N, int K, long long ldw,
                                   int kchunk) {
  constexpr int NPT = 4;            // cols per thread
  const int tid = threadIdx.x;
  const int n0 = blockIdx.x * 256 * NPT + tid * NPT;
  const int ks = blockIdx.y;
  const int k0 = ks * kchunk;
  const int k1 = min(K, k0 + kchunk);
  __shared__ short x_lds[MT * 128];     // x chunk [M][128]

  float acc[MT][NPT];
#pragma unroll
  for (int m = 0; m < MT; ++m)
#pragma unroll
    for (int j = 0; j < NPT; ++j) acc[m][j] = 0.f;

  const bool nok = n0 + NPT <= N;
  for (int kb = k0; kb < k1; kb += 128) {
    const int kc = min(128, k1 - kb);
    // stage x[M][kc] (tiny): 256 threads x 2 elems covers 32x128
    for (int i = tid * 2; i < MT * 128; i += 512) {
      int m = i / 128, kk = i % 128;
      short v0 = 0, v1 = 0;
      if (m < M && kk < kc) v0 = xg[(long long)m * K + kb + kk];
      if (m < M && kk + 1 < kc) v1 = xg[(long long)m * K + kb + kk + 1];
      x_lds[m * 128 + kk] = v0;
      if (kk + 1 < 128) x_lds[m * 128 + kk + 1] = v1;
    }
    __syncthreads();
    if (nok) {
      for (int kk = 0; kk < kc; ++kk) {
        shortx4 w4 = *reinterpret_cast<const shortx4*>(
            wg + (long long)(kb + kk) * ldw + n0);
        float wf[NPT];
#pragma unroll
        for (int j = 0; j < NPT; ++j) wf[j] = bf2f(w4[j]);
#pragma unroll
        for (int m = 0; m < MT; ++m) {
          float xv = bf2f(x_lds[m * 128 + kk]);
#pragma unroll
          for (int j = 0; j < NPT; ++j) acc[m][j] += xv * wf[j];
        }
      }
    } else if (n0 < N) {
      for (int kk = 0; kk < kc; ++kk) {
#pragma unroll
        for (int j = 0; j < NPT; ++j) {
          if (n0 + j >= N) break;
          float wf = bf2f(wg[(long long)(kb + kk) * ldw + n0 + j]);
#pragma unroll
          for (int m = 0; m < MT; ++m)
            acc[m][j] += bf2f(x_lds[m * 128 + kk]) * wf;
        }
      }
    }
    __syncthreads();
  }
  if (n0 >= N) return;
  float* out = partial + ((long long)ks * MT) * N;
#pragma unroll
  for (int m = 0; m < MT; ++m)
#pragma unroll
    for (int j = 0; j < NPT; ++j)
      if (n0 + j < N) out[(long long)m * N + n0 + j] = acc[m][j];
}

// reduce over ksplit + bias + bf16 cast: y[M,N]
template <int DT>
__global__ void decode_gemm_reduce_kernel(const float* __restrict__ partial,
                                          const short* __restrict__ bias,
                                          short* __restrict__ y, int M, int N,
                                          int mt, int ksplit,
                                          const float* __restrict__ chscale = nullptr) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = (long long)M * N;
  if (i >= total) return;
  int m = (int)(i / N), n = (int)(i % N);
  float acc = 0.f;
  for (int s = 0; s < ksplit; ++s)
    acc += partial[((long long)s * mt + m) * N + n];
  if (chscale) acc *= chscale[n] * (1.f / 127.f);
  if (bias) acc += bf2f(bias[n]);
  y[i] = f2bf(acc);
}

void decode_gemm(const void* x, const void* w, const void* bias, void* y,
                 float* workspace, int64_t m, int64_t n, int64_t k,
                 int64_t ldw, int64_t ksplit, hipStream_t s) {
  const int mt = m <= 8 ? 8 : (m <= 16 ? 16 : 32);
  const int kchunk = (int)((k + ksplit - 1) / ksplit + 127) / 128 * 128;
  const int ks = (int)((k + kchunk - 1) / kchunk);
  dim3 grid((unsigned)((n + 1023) / 1024), (unsigned)ks);
#define DG(MT)                                                              \
  hipLaunchKernelGGL((decode_gemm_kernel<MT>), grid, dim3(256), 0, s,       \
                     (const short*)x, (const short*)w, workspace, (int)m,   \
                     (int)n, (int)k, ldw, kchunk)
  if (mt == 8) DG(8);
  else if (mt == 16) DG(16);
  else DG(32);
#undef DG
  long long total = m * n;
  dim3 rg((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL((decode_gemm_reduce_kernel<0>), rg, dim3(256), 0, s,
                     workspace, (const short*)bias, (short*)y, (int)m, (int)n,
                     mt, ks);
}

}  // namespace pa

// ---------------------------------------------------------------------------
// MFMA W-streaming decode GEMM (round-2 rework of the split-K streamer
// above: the vector-ALU inner product was VALU-bound at M=32; MFMA makes
// the arithmetic free so the kernel can run at the weight-read roofline).
//
//   y[M,N] = x[M,K] @ W[K,N],  M <= 32, N % 256 == 0, K % 64 == 0
//
// W rows are read 16 B/lane coalesced and TRANSPOSED through LDS
// (wt[n][k] rows) so the mfma_f32_16x16x32_bf16 B-fragment (8 consecutive
// k at fixed n -- layout verified by probe.hip) is one ds_read_b128.
// x is tiny and L2-resident: A-fragments load straight from global.
// Split-K across grid.y writes fp32 partials reduced by
// decode_gemm_reduce_kernel.
// ---------------------------------------------------------------------------
namespace pa {

// k-chunk swizzle: chunk kc (8 bf16) of row n lives at kc ^ ((n>>3)&7)
__device__ __forceinline__ int dg_swz(int kc, int n) {
  return (kc ^ ((n >> 3) & 7)) * 8;
}

typedef __attribute__((ext_vector_type(8))) signed char scharx8;
typedef __attribute__((ext_vector_type(16))) signed char scharx16;

// INT8: W is [K, N] int8 (weight_quantize layout); values convert to
// bf16 in-register before LDS staging so the MFMA path is unchanged --
// the per-channel scale/127 applies in the reduce kernel.
template <int MTILES, bool INT8 = false>  // MTILES 1: M<=16, 2: M<=32
__launch_bounds__(256, 2)
__global__ void decode_gemm_mfma_kernel(const short* __restrict__ xg,
                                        const short* __restrict__ wg,
                                        float* __restrict__ partial,
                                        int M, int N, int K, long long ldw,
                                        int kchunk) {
  constexpr int NB = 256;       // n cols per block
  constexpr int KT = 64;        // k rows per LDS tile
  __shared__ short wt[2][NB * KT];

  const int n0 = blockIdx.x * NB;
  const int ks = blockIdx.y;
  const int k0 = ks * kchunk;
  const int k1 = min(K, k0 + kchunk);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;      // wave: owns n cols [wv*64, wv*64+64)
  const int l16 = lane & 15;
  const int lg = lane >> 4;

  // staging map (bf16): 8 rounds x (256 thr x 8 elem); thread covers k
  // row (r*8 + tid/32) at n segment (tid%32)*8.
  // staging map (int8): 4 rounds x (256 thr x 16 elem) -- 16B loads keep
  // the same bytes in flight as the bf16 path (8B loads halved the
  // outstanding bytes and left the kernel latency-bound at ~1 TB/s).
  // int8 map: 2 rounds x (k-pair, 16 n) per thread -- adjacent k rows
  // pack into ONE b32 LDS write per n (b16 scatters were the bound:
  // the int8 tile has half the bytes but the same element count)
  const int s_k = INT8 ? (tid >> 4) * 2 : (tid >> 5);
  const int s_n = INT8 ? (tid & 15) * 16 : (tid & 31) * 8;

  floatx4 acc[MTILES][4];
#pragma unroll
  for (int mt = 0; mt < MTILES; ++mt)
#pragma unroll
    for (int f = 0; f < 4; ++f)
      acc[mt][f] = floatx4{0.f, 0.f, 0.f, 0.f};

  // x A-fragment rows for this lane (row index == l16 within each m-tile)
  bool xrow_ok[MTILES];
  long long xbase[MTILES];
#pragma unroll
  for (int mt = 0; mt < MTILES; ++mt) {
    int row = mt * 16 + l16;
    xrow_ok[mt] = row < M;
    xbase[mt] = (long long)row * K;
  }

  // INT8 keeps raw 16B int8 loads in TWO reg sets (prefetch depth 2 --
  // one tile of int8 is only half the bytes of a bf16 tile, and depth-1
  // prefetch left the kernel in-flight-limited at ~1.2 TB/s).  The sets
  // are NAMED arrays selected at compile time via distinct call sites
  // (a runtime slot index into a register array spills to scratch,
  // rule #20: measured 960 us vs 50 us).  bf16->LDS conversion happens
  // at write time.
  shortx8 stg[8];
  scharx16 qstgA[4], qstgB[4];
  const signed char* w8 = reinterpret_cast<const signed char*>(wg);
  auto load_q = [&](scharx16(&q)[4], int kb) {
#pragma unroll
    for (int r = 0; r < 2; ++r)
#pragma unroll
      for (int p = 0; p < 2; ++p) {
        int k = kb + r * 32 + s_k + p;
        if (k >= k1) {
#pragma unroll
          for (int j = 0; j < 16; ++j) q[2 * r + p][j] = 0;
        } else {
          q[2 * r + p] = *reinterpret_cast<const scharx16*>(
              w8 + (long long)k * ldw + n0 + s_n);
        }
      }
  };
  auto write_q = [&](const scharx16(&q)[4], int buf) {
#pragma unroll
    for (int r = 0; r < 2; ++r) {
      int kl = r * 32 + s_k;        // even; kl and kl+1 share a chunk
      int kc = kl >> 3, ko = kl & 7;
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        int n = s_n + j;
        unsigned lo = (unsigned short)f2bf((float)q[2 * r][j]);
        unsigned hi = (unsigned short)f2bf((float)q[2 * r + 1][j]);
        *reinterpret_cast<unsigned int*>(
            &wt[buf][n * KT + dg_swz(kc, n) + ko]) = lo | (hi << 16);
      }
    }
  };
  auto load_tile = [&](int kb) {    // bf16 path: global -> regs
#pragma unroll
    for (int r = 0; r < 8; ++r) {
      int k = kb + r * 8 + s_k;
      stg[r] = (k < k1)
          ? *reinterpret_cast<const shortx8*>(wg + (long long)k * ldw + n0 + s_n)
          : shortx8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  };
  auto write_tile = [&](int buf) {  // bf16 path: regs -> LDS transposed
#pragma unroll
    for (int r = 0; r < 8; ++r) {
      int kl = r * 8 + s_k;         // k row within tile
      int kc = kl >> 3, ko = kl & 7;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        wt[buf][(s_n + j) * KT + dg_swz(kc, s_n + j) + ko] = stg[r][j];
    }
  };

  auto load_a = [&](shortx8 af[][2], int kb) {
#pragma unroll
    for (int mt = 0; mt < MTILES; ++mt)
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
        int k = kb + kt * 32 + lg * 8;
        af[mt][kt] = (xrow_ok[mt] && k + 7 < K)
            ? *reinterpret_cast<const shortx8*>(xg + xbase[mt] + k)
            : shortx8{0, 0, 0, 0, 0, 0, 0, 0};
      }
  };

  // One barrier per tile: at iter t the regs hold tile t+1 (loaded at
  // t-1), which is written into buf (t+1)%2 BEFORE computing tile t from
  // buf t%2.  WAR on buf t%2 (written next iter) and RAW on buf (t+1)%2
  // (read next iter) are both cut by the single end-of-iter barrier.
  shortx8 af[MTILES][2], afn[MTILES][2];
  if (INT8) {
    load_q(qstgA, k0);
    write_q(qstgA, 0);
    load_q(qstgA, k0 + KT);        // A = tile 1
    load_q(qstgB, k0 + 2 * KT);    // B = tile 2 (depth-2 prefetch)
  } else {
    load_tile(k0);
    write_tile(0);
    load_tile(k0 + KT);            // regs = tile 1
  }
  load_a(af, k0);
  __syncthreads();

  int cur = 0;
  bool useA = true;
  for (int kb = k0; kb < k1; kb += KT) {
    const bool more = kb + KT < k1;
    if (INT8) {
      if (more) {
        if (useA) write_q(qstgA, cur ^ 1);
        else      write_q(qstgB, cur ^ 1);
      }
      if (kb + 3 * KT < k1) {
        if (useA) load_q(qstgA, kb + 3 * KT);
        else      load_q(qstgB, kb + 3 * KT);
      }
      useA = !useA;
    } else {
      if (more) write_tile(cur ^ 1);  // regs from two iters back -> next buf
      if (kb + 2 * KT < k1) load_tile(kb + 2 * KT);
    }
    if (more) load_a(afn, kb + KT);
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      int kc = kt * 4 + lg;
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        int n = wv * 64 + f * 16 + l16;
        shortx8 bf = *reinterpret_cast<const shortx8*>(
            &wt[cur][n * KT + dg_swz(kc, n)]);
#pragma unroll
        for (int mt = 0; mt < MTILES; ++mt)
          acc[mt][f] = mfma_bf16(af[mt][kt], bf, acc[mt][f]);
      }
    }
#pragma unroll
    for (int mt = 0; mt < MTILES; ++mt)
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) af[mt][kt] = afn[mt][kt];
    cur ^= 1;
    __syncthreads();
  }

  // partial[ks][mt_pad][N] fp32; C lane layout: row lg*4+r, col l16
  const int mt_pad = MTILES * 16;
  float* out = partial + (long long)ks * mt_pad * N;
#pragma unroll
  for (int mt = 0; mt < MTILES; ++mt)
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int n = n0 + wv * 64 + f * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = mt * 16 + lg * 4 + r;
        out[(long long)row * N + n] = acc[mt][f][r];
      }
    }
}

void decode_gemm_mfma(const void* x, const void* w, const void* bias, void* y,
                      float* workspace, int64_t m, int64_t n, int64_t k,
                      int64_t ldw, int64_t ksplit, hipStream_t s,
                      const float* chscale, bool int8w) {
  const int mt = m <= 16 ? 16 : 32;
  // kchunk: multiple of 64 so every split starts tile-aligned
  int kchunk = (int)(((k + ksplit - 1) / ksplit + 63) / 64) * 64;
  const int ks = (int)((k + kchunk - 1) / kchunk);
  dim3 grid((unsigned)(n / 256), (unsigned)ks);
#define DGM(MT, I8)                                                          \
  hipLaunchKernelGGL((decode_gemm_mfma_kernel<MT, I8>), grid, dim3(256), 0,  \
                     s, (const short*)x, (const short*)w, workspace, (int)m, \
                     (int)n, (int)k, ldw, kchunk)
  if (int8w) { if (mt == 16) DGM(1, true); else DGM(2, true); }
  else       { if (mt == 16) DGM(1, false); else DGM(2, false); }
#undef DGM
  long long total = m * n;
  dim3 rg((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL((decode_gemm_reduce_kernel<0>), rg, dim3(256), 0, s,
                     workspace, (const short*)bias, (short*)y, (int)m, (int)n,
                     mt, ks, chscale);
}

}  // namespace pa
