// Fused AdamW on flat parameter shards + grad-norm kernel.
//
// MI355X design: the sharded optimizer (ZeRO) keeps ONE contiguous fp32
// master/m/v buffer per rank and a bf16 model-weight shard; this kernel
// walks all of it in a single launch (4 float4 streams in, 2 out),
// HBM-bound at ~8 TB/s.  Replaces the reference's per-param
// adamw_kernel.cu + multi_tensor_apply.h (fused_adam_kernel.cu) -- the
// flat-shard layout makes multi-tensor machinery unnecessary.
#include "common.h"
#include "api.h"

namespace pa {

template <int GDT, bool BF16OUT>
__global__ void adamw_kernel(float* __restrict__ master, void* __restrict__ param_bf16,
                             const void* __restrict__ grad, float* __restrict__ m,
                             float* __restrict__ v, int64_t numel, float lr,
                             float beta1, float beta2, float eps, float wd,
                             float bias1, float bias2, float gscale) {
  // bias1 = 1 - beta1^t, bias2 = 1 - beta2^t
  const float inv_b1 = 1.f / bias1;
  const float inv_b2 = 1.f / bias2;
  const int64_t numel4 = numel & ~3ll;  // float4 body; <4 tail handled scalar below
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4; i < numel4;
       i += (int64_t)gridDim.x * blockDim.x * 4) {
    float4 pm = *reinterpret_cast<float4*>(master + i);
    float4 mm = *reinterpret_cast<float4*>(m + i);
    float4 vv = *reinterpret_cast<float4*>(v + i);
    float g[4];
    if (GDT == kBF16) {
      shortx4 gv = *reinterpret_cast<const shortx4*>((const short*)grad + i);
#pragma unroll
      for (int k = 0; k < 4; ++k) g[k] = bf2f(gv[k]) * gscale;
    } else {
      float4 gv = *reinterpret_cast<const float4*>((const float*)grad + i);
      g[0] = gv.x * gscale; g[1] = gv.y * gscale; g[2] = gv.z * gscale; g[3] = gv.w * gscale;
    }
    float p[4] = {pm.x, pm.y, pm.z, pm.w};
    float mo[4] = {mm.x, mm.y, mm.z, mm.w};
    float vo[4] = {vv.x, vv.y, vv.z, vv.w};
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      // decoupled weight decay (AdamW): p *= (1 - lr*wd)
      p[k] -= lr * wd * p[k];
      mo[k] = beta1 * mo[k] + (1.f - beta1) * g[k];
      vo[k] = beta2 * vo[k] + (1.f - beta2) * g[k] * g[k];
      float mhat = mo[k] * inv_b1;
      float vhat = vo[k] * inv_b2;
      p[k] -= lr * mhat / (sqrtf(vhat) + eps);
    }
    *reinterpret_cast<float4*>(master + i) = make_float4(p[0], p[1], p[2], p[3]);
    *reinterpret_cast<float4*>(m + i) = make_float4(mo[0], mo[1], mo[2], mo[3]);
    *reinterpret_cast<float4*>(v + i) = make_float4(vo[0], vo[1], vo[2], vo[3]);
    if (BF16OUT) {
      shortx4 pv;
#pragma unroll
      for (int k = 0; k < 4; ++k) pv[k] = f2bf(p[k]);
      *reinterpret_cast<shortx4*>((short*)param_bf16 + i) = pv;
    } else if (param_bf16) {
      *reinterpret_cast<float4*>((float*)param_bf16 + i) = make_float4(p[0], p[1], p[2], p[3]);
    }
  }
  for (int64_t i = numel4 + (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < numel;
       i += (int64_t)gridDim.x * blockDim.x) {
    float g = (GDT == kBF16) ? bf2f(((const short*)grad)[i]) * gscale
                             : ((const float*)grad)[i] * gscale;
    float p = master[i];
    p -= lr * wd * p;
    float mo = beta1 * m[i] + (1.f - beta1) * g;
    float vo = beta2 * v[i] + (1.f - beta2) * g * g;
    p -= lr * (mo * inv_b1) / (sqrtf(vo * inv_b2) + eps);
    master[i] = p; m[i] = mo; v[i] = vo;
    if (BF16OUT) ((short*)param_bf16)[i] = f2bf(p);
    else if (param_bf16) ((float*)param_bf16)[i] = p;
  }
}

void adamw(float* master, void* param_bf16, const void* grad, float* m,
           float* v, int64_t numel, float lr, float beta1, float beta2,
           float eps, float wd, float beta1_pow, float beta2_pow,
           float grad_scale, int grad_dtype, bool param_out_bf16, hipStream_t s) {
  float bias1 = 1.f - beta1_pow;
  float bias2 = 1.f - beta2_pow;
  dim3 g((unsigned)elementwise_grid(cdiv((int)hmin<int64_t>(numel / 8 + 1, 1 << 30), 256)));
  if (grad_dtype == kBF16) {
    if (param_out_bf16)
      hipLaunchKernelGGL((adamw_kernel<kBF16, true>), g, dim3(256), 0, s, master,
                         param_bf16, grad, m, v, numel, lr, beta1, beta2, eps, wd, bias1, bias2, grad_scale);
    else
      hipLaunchKernelGGL((adamw_kernel<kBF16, false>), g, dim3(256), 0, s, master,
                         param_bf16, grad, m, v, numel, lr, beta1, beta2, eps, wd, bias1, bias2, grad_scale);
  } else {
    if (param_out_bf16)
      hipLaunchKernelGGL((adamw_kernel<kF32, true>), g, dim3(256), 0, s, master,
                         param_bf16, grad, m, v, numel, lr, beta1, beta2, eps, wd, bias1, bias2, grad_scale);
    else
      hipLaunchKernelGGL((adamw_kernel<kF32, false>), g, dim3(256), 0, s, master,
                         param_bf16, grad, m, v, numel, lr, beta1, beta2, eps, wd, bias1, bias2, grad_scale);
  }
}

template <int DT>
__global__ void l2norm_sq_kernel(const void* __restrict__ x, float* __restrict__ out,
                                 int64_t numel) {
  __shared__ float red[4];
  float acc = 0.f;
  const int64_t numel4 = numel & ~3ll;  // float4 body; <4 tail handled scalar below
  int64_t per_blk = ((numel4 / 4 + gridDim.x - 1) / gridDim.x) * 4;
  int64_t blk1 = min(numel4, (int64_t)(blockIdx.x + 1) * per_blk);
  for (int64_t i = (int64_t)blockIdx.x * per_blk + (int64_t)threadIdx.x * 4; i < blk1;
       i += (int64_t)blockDim.x * 4) {
    if (DT == kBF16) {
      shortx4 v = *reinterpret_cast<const shortx4*>((const short*)x + i);
#pragma unroll
      for (int k = 0; k < 4; ++k) { float f = bf2f(v[k]); acc += f * f; }
    } else {
      float4 v = *reinterpret_cast<const float4*>((const float*)x + i);
      acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
  }
  if (blockIdx.x == 0) {
    for (int64_t i = numel4 + threadIdx.x; i < numel; i += blockDim.x) {
      float f = (DT == kBF16) ? bf2f(((const short*)x)[i]) : ((const float*)x)[i];
      acc += f * f;
    }
  }
  float t = block_reduce_256(acc, SumOp(), red, 0.f);
  if (threadIdx.x == 0) atomicAdd(out, t);
}

void l2norm_sq(const void* x, float* out, int64_t numel, int dtype, hipStream_t s) {
  dim3 g((unsigned)elementwise_grid(cdiv((int)hmin<int64_t>(numel / 4 + 1, 1 << 30), 256)));
  if (dtype == kBF16)
    hipLaunchKernelGGL((l2norm_sq_kernel<kBF16>), g, dim3(256), 0, s, x, out, numel);
  else
    hipLaunchKernelGGL((l2norm_sq_kernel<kF32>), g, dim3(256), 0, s, x, out, numel);
}

}  // namespace pa
