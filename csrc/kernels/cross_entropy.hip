// Fused softmax + cross-entropy (hard labels) for gfx950.
//
// One 256-thread block per row, online (max,sum) accumulation in a single
// pass over V (vocab ~50k: 2 passes over HBM total for fwd incl. store of
// nothing -- we keep only lse + loss, grads recompute exp in bwd).
//
// Reference behavior parity: paddle/phi/kernels/gpu/cross_entropy_kernel.cu
// (VectorizedSoftmaxForward / WarpSoftmaxForward) -- re-derived for wave64.
#include "common.h"
#include "api.h"

namespace pa {

template <int DT> struct VIO2;
template <> struct VIO2<kBF16> {
  static __device__ __forceinline__ void load8(const void* p, int64_t idx, float* f) {
    shortx8 v = *reinterpret_cast<const shortx8*>((const short*)p + idx);
#pragma unroll
    for (int i = 0; i < 8; ++i) f[i] = bf2f(v[i]);
  }
  static __device__ __forceinline__ void store8(void* p, int64_t idx, const float* f) {
    shortx8 v;
#pragma unroll
    for (int i = 0; i < 8; ++i) v[i] = f2bf(f[i]);
    *reinterpret_cast<shortx8*>((short*)p + idx) = v;
  }
  static __device__ __forceinline__ float load1(const void* p, int64_t idx) {
    return bf2f(((const short*)p)[idx]);
  }
  static __device__ __forceinline__ void store1(void* p, int64_t idx, float f) {
    ((short*)p)[idx] = f2bf(f);
  }
};
template <> struct VIO2<kF32> {
  static __device__ __forceinline__ void load8(const void* p, int64_t idx, float* f) {
    const float4* q = reinterpret_cast<const float4*>((const float*)p + idx);
    float4 a = q[0], b = q[1];
    f[0]=a.x; f[1]=a.y; f[2]=a.z; f[3]=a.w; f[4]=b.x; f[5]=b.y; f[6]=b.z; f[7]=b.w;
  }
  static __device__ __forceinline__ void store8(void* p, int64_t idx, const float* f) {
    float4* q = reinterpret_cast<float4*>((float*)p + idx);
    q[0] = make_float4(f[0],f[1],f[2],f[3]); q[1] = make_float4(f[4],f[5],f[6],f[7]);
  }
  static __device__ __forceinline__ float load1(const void* p, int64_t idx) {
    return ((const float*)p)[idx];
  }
  static __device__ __forceinline__ void store1(void* p, int64_t idx, float f) {
    ((float*)p)[idx] = f;
  }
};

// merge two (max, sumexp) pairs
__device__ __forceinline__ void lse_merge(float& m, float& l, float m2, float l2) {
  float mn = fmaxf(m, m2);
  // guard -inf - -inf
  float a = (m == mn) ? l : l * __expf(m - mn);
  float b = (m2 == mn) ? l2 : l2 * __expf(m2 - mn);
  if (m == -INFINITY && m2 == -INFINITY) { l = 0.f; m = -INFINITY; return; }
  l = a + b;
  m = mn;
}

template <int DT>
__global__ void ce_fwd_kernel(const void* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              float* __restrict__ loss, float* __restrict__ lse_out,
                              int64_t n, int64_t v, int64_t ignore_index) {
  __shared__ float red_m[4], red_l[4];
  for (int64_t row = blockIdx.x; row < n; row += gridDim.x) {
    const int64_t base = row * v;
    float m = -INFINITY, l = 0.f;
    int64_t i = threadIdx.x * 8;
    const int64_t v8 = v & ~7LL;
    for (; i < v8; i += blockDim.x * 8) {
      float f[8];
      VIO2<DT>::load8(logits, base + i, f);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float x = f[k];
        if (x > m) { l *= __expf(m - x); m = x; }
        l += __expf(x - m);
      }
    }
    // tail
    for (int64_t j = v8 + threadIdx.x; j < v; j += blockDim.x) {
      float x = VIO2<DT>::load1(logits, base + j);
      if (x > m) { l *= __expf(m - x); m = x; }
      l += __expf(x - m);
    }
    // wave reduce
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float m2 = __shfl_xor(m, off, WAVE);
      float l2 = __shfl_xor(l, off, WAVE);
      lse_merge(m, l, m2, l2);
    }
    int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
    if (lane == 0) { red_m[wid] = m; red_l[wid] = l; }
    __syncthreads();
    m = red_m[0]; l = red_l[0];
#pragma unroll
    for (int k = 1; k < 4; ++k) lse_merge(m, l, red_m[k], red_l[k]);
    float lse = m + __logf(l);
    if (threadIdx.x == 0) {
      int64_t lab = labels[row];
      lse_out[row] = lse;
      if (lab == ignore_index) {
        loss[row] = 0.f;
      } else {
        float xl = VIO2<DT>::load1(logits, base + lab);
        loss[row] = lse - xl;
      }
    }
    __syncthreads();
  }
}

template <int DT>
__global__ void ce_bwd_kernel(const float* __restrict__ dloss,
                              const void* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              const float* __restrict__ lse, void* __restrict__ dlogits,
                              int64_t n, int64_t v, int64_t ignore_index) {
  for (int64_t row = blockIdx.x; row < n; row += gridDim.x) {
    const int64_t base = row * v;
    const int64_t lab = labels[row];
    const float g = (lab == ignore_index) ? 0.f : dloss[row];
    const float ls = lse[row];
    const int64_t v8 = v & ~7LL;
    int64_t i = threadIdx.x * 8;
    for (; i < v8; i += blockDim.x * 8) {
      float f[8];
      VIO2<DT>::load8(logits, base + i, f);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float p = __expf(f[k] - ls);
        f[k] = g * (p - ((i + k) == lab ? 1.f : 0.f));
      }
      VIO2<DT>::store8(dlogits, base + i, f);
    }
    for (int64_t j = v8 + threadIdx.x; j < v; j += blockDim.x) {
      float p = __expf(VIO2<DT>::load1(logits, base + j) - ls);
      VIO2<DT>::store1(dlogits, base + j, g * (p - (j == lab ? 1.f : 0.f)));
    }
  }
}

void softmax_ce_fwd(const void* logits, const int64_t* labels, float* loss,
                    float* lse, int64_t n, int64_t v, int64_t ignore_index,
                    int dtype, hipStream_t s) {
  int grid = (int)(n < 2048 ? n : 2048);
  if (dtype == kBF16)
    hipLaunchKernelGGL((ce_fwd_kernel<kBF16>), dim3(grid), dim3(256), 0, s,
                       logits, labels, loss, lse, n, v, ignore_index);
  else
    hipLaunchKernelGGL((ce_fwd_kernel<kF32>), dim3(grid), dim3(256), 0, s,
                       logits, labels, loss, lse, n, v, ignore_index);
}

void softmax_ce_bwd(const float* dloss, const void* logits,
                    const int64_t* labels, const float* lse, void* dlogits,
                    int64_t n, int64_t v, int64_t ignore_index, int dtype,
                    hipStream_t s) {
  int grid = (int)(n < 2048 ? n : 2048);
  if (dtype == kBF16)
    hipLaunchKernelGGL((ce_bwd_kernel<kBF16>), dim3(grid), dim3(256), 0, s,
                       dloss, logits, labels, lse, dlogits, n, v, ignore_index);
  else
    hipLaunchKernelGGL((ce_bwd_kernel<kF32>), dim3(grid), dim3(256), 0, s,
                       dloss, logits, labels, lse, dlogits, n, v, ignore_index);
}

}  // namespace pa
