// Weight-only int8 GEMV for decode-time linears.
//
// Serving decode is memory-bound on weight reads; int8 weights halve the
// bytes vs bf16.  Layout: qweight [N, K] row-major int8 (one contiguous
// row per output channel -- sequential HBM streams), scale [N] fp32,
// out[m,n] = sum_k x[m,k] * qw[n,k] * scale[n] / 127 (+ bias[n]).
// One wave per output channel; lanes stride k with 8-byte int8 loads.
// Reference behavior: paddle/phi/kernels/funcs/weight_only_gemv.cu,
// weight_quantize_kernel.cu (re-derived for wave64).
#include "common.h"
#include "api.h"

namespace pa {

typedef __attribute__((ext_vector_type(8))) char charx8;

template <int DT> struct WLS;
template <> struct WLS<kBF16> {
  static __device__ __forceinline__ void load8(const void* p, int64_t i, float* f) {
    shortx8 v = *reinterpret_cast<const shortx8*>((const short*)p + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) f[j] = bf2f(v[j]);
  }
};
template <> struct WLS<kF32> {
  static __device__ __forceinline__ void load8(const void* p, int64_t i, float* f) {
    const float4* q = reinterpret_cast<const float4*>((const float*)p + i);
    float4 a = q[0], b = q[1];
    f[0]=a.x; f[1]=a.y; f[2]=a.z; f[3]=a.w; f[4]=b.x; f[5]=b.y; f[6]=b.z; f[7]=b.w;
  }
};

template <int DT>
__global__ void wo_gemv_kernel(const void* __restrict__ x,
                               const signed char* __restrict__ wq,
                               const float* __restrict__ scale,
                               const void* __restrict__ bias,
                               void* __restrict__ out,
                               int M, int N, int64_t K) {
  const int wv = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = blockIdx.x * (blockDim.x >> 6) + wv;
  const int m = blockIdx.y;
  if (n >= N) return;
  const signed char* wr = wq + (int64_t)n * K;
  float acc = 0.f;
  for (int64_t k = (int64_t)lane * 8; k < K; k += 64 * 8) {
    charx8 wv8 = *reinterpret_cast<const charx8*>(wr + k);
    float xf[8];
    WLS<DT>::load8(x, (int64_t)m * K + k, xf);
#pragma unroll
    for (int i = 0; i < 8; ++i) acc += xf[i] * (float)wv8[i];
  }
  acc = wave_reduce(acc, SumOp());
  if (lane == 0) {
    float v = acc * scale[n] * (1.f / 127.f);
    if (bias) v += (DT == kBF16) ? bf2f(((const short*)bias)[n])
                                 : ((const float*)bias)[n];
    if (DT == kBF16) ((short*)out)[(int64_t)m * N + n] = f2bf(v);
    else ((float*)out)[(int64_t)m * N + n] = v;
  }
}

void weight_only_gemv(const void* x, const void* wq, const float* scale,
                      const void* bias, void* out, int64_t m, int64_t n,
                      int64_t k, int dtype, hipStream_t s) {
  dim3 grid((unsigned)cdiv((int)n, 4), (unsigned)m);
  if (dtype == kBF16)
    hipLaunchKernelGGL((wo_gemv_kernel<kBF16>), grid, dim3(256), 0, s, x,
                       (const signed char*)wq, scale, bias, out, (int)m, (int)n, k);
  else
    hipLaunchKernelGGL((wo_gemv_kernel<kF32>), grid, dim3(256), 0, s, x,
                       (const signed char*)wq, scale, bias, out, (int)m, (int)n, k);
}

}  // namespace pa
