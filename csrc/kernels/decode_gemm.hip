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
                                          int mt, int ksplit) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = (long long)M * N;
  if (i >= total) return;
  int m = (int)(i / N), n = (int)(i % N);
  float acc = 0.f;
  for (int s = 0; s < ksplit; ++s)
    acc += partial[((long long)s * mt + m) * N + n];
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
