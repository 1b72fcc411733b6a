// Embedding lookup + scatter-add gradient for gfx950.
//
// Reference behavior parity: paddle/phi/kernels/gpu/embedding_grad_kernel.cu
// (EmbeddingGrad, atomic path).  fwd: one wave per token row; bwd: fp32
// atomics into the dense grad table (device-scope atomicAdd, guide G12).
#include "common.h"
#include "api.h"

namespace pa {

template <int DT>
__global__ void embedding_fwd_kernel(const void* __restrict__ table,
                                     const int64_t* __restrict__ ids,
                                     void* __restrict__ out, int64_t n_ids,
                                     int64_t d, int64_t vocab, int64_t padding_idx) {
  int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int lane = threadIdx.x & 63;
  int64_t nwaves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  for (int64_t r = wid; r < n_ids; r += nwaves) {
    int64_t id = ids[r];
    bool pad = (id == padding_idx) || id < 0 || id >= vocab;
    for (int64_t j = lane * 8; j < d; j += 64 * 8) {
      if (DT == kBF16) {
        shortx8 v;
        if (pad) { for (int k = 0; k < 8; ++k) v[k] = 0; }
        else v = *reinterpret_cast<const shortx8*>((const short*)table + id * d + j);
        *reinterpret_cast<shortx8*>((short*)out + r * d + j) = v;
      } else {
        float4 a = {0,0,0,0}, b = {0,0,0,0};
        if (!pad) {
          const float4* p = reinterpret_cast<const float4*>((const float*)table + id * d + j);
          a = p[0]; b = p[1];
        }
        float4* q = reinterpret_cast<float4*>((float*)out + r * d + j);
        q[0] = a; q[1] = b;
      }
    }
  }
}

template <int DT>
__global__ void embedding_bwd_kernel(const void* __restrict__ dout,
                                     const int64_t* __restrict__ ids,
                                     float* __restrict__ dtable, int64_t n_ids,
                                     int64_t d, int64_t vocab, int64_t padding_idx) {
  int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int lane = threadIdx.x & 63;
  int64_t nwaves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  for (int64_t r = wid; r < n_ids; r += nwaves) {
    int64_t id = ids[r];
    if (id == padding_idx || id < 0 || id >= vocab) continue;
    for (int64_t j = lane * 2; j < d; j += 64 * 2) {
      float f0, f1;
      if (DT == kBF16) {
        const short* p = (const short*)dout + r * d + j;
        f0 = bf2f(p[0]); f1 = bf2f(p[1]);
      } else {
        const float* p = (const float*)dout + r * d + j;
        f0 = p[0]; f1 = p[1];
      }
      atomicAdd(&dtable[id * d + j], f0);
      atomicAdd(&dtable[id * d + j + 1], f1);
    }
  }
}

void embedding_fwd(const void* table, const int64_t* ids, void* out,
                   int64_t n_ids, int64_t d, int64_t vocab, int64_t padding_idx,
                   int dtype, hipStream_t s) {
  dim3 g((unsigned)hmin<int64_t>(cdiv((int)hmin<int64_t>(n_ids, 1 << 24), 4), 2048));
  if (dtype == kBF16)
    hipLaunchKernelGGL((embedding_fwd_kernel<kBF16>), g, dim3(256), 0, s, table, ids, out, n_ids, d, vocab, padding_idx);
  else
    hipLaunchKernelGGL((embedding_fwd_kernel<kF32>), g, dim3(256), 0, s, table, ids, out, n_ids, d, vocab, padding_idx);
}

void embedding_bwd(const void* dout, const int64_t* ids, float* dtable,
                   int64_t n_ids, int64_t d, int64_t vocab, int64_t padding_idx,
                   int dtype, hipStream_t s) {
  dim3 g((unsigned)hmin<int64_t>(cdiv((int)hmin<int64_t>(n_ids, 1 << 24), 4), 2048));
  if (dtype == kBF16)
    hipLaunchKernelGGL((embedding_bwd_kernel<kBF16>), g, dim3(256), 0, s, dout, ids, dtable, n_ids, d, vocab, padding_idx);
  else
    hipLaunchKernelGGL((embedding_bwd_kernel<kF32>), g, dim3(256), 0, s, dout, ids, dtable, n_ids, d, vocab, padding_idx);
}

}  // namespace pa
