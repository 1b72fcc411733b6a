// Fused elementwise kernels: bias+gelu, swiglu, rope, dropout+residual,
// column-sum -- all HBM-bound, vectorized 8-wide, grid-stride.
//
// Reference behavior parity: paddle/phi/kernels/fusion/gpu/
// fused_bias_act_kernel.cu, fused_dropout_add_kernel.cu,
// fused_rope_kernel.cu (RotateHalf) -- re-derived for wave64/CDNA4
// (guide Appendix B: trig tables precomputed on host, never on-device).
#include "common.h"
#include "api.h"

namespace pa {

template <int DT> struct LS8;
template <> struct LS8<kBF16> {
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
};
template <> struct LS8<kF32> {
  static __device__ __forceinline__ void load8(const void* p, int64_t idx, float* f) {
    const float4* q = reinterpret_cast<const float4*>((const float*)p + idx);
    float4 a = q[0], b = q[1];
    f[0]=a.x; f[1]=a.y; f[2]=a.z; f[3]=a.w; f[4]=b.x; f[5]=b.y; f[6]=b.z; f[7]=b.w;
  }
  static __device__ __forceinline__ void store8(void* p, int64_t idx, const float* f) {
    float4* q = reinterpret_cast<float4*>((float*)p + idx);
    q[0] = make_float4(f[0],f[1],f[2],f[3]); q[1] = make_float4(f[4],f[5],f[6],f[7]);
  }
};

// erf via Abramowitz-Stegun 7.1.26 (|err| < 1.5e-7, far below bf16 ulp):
// libm erff expands to ~60 VALU ops with branch blends, which made the
// fused bias+gelu kernel VALU-bound (measured 3.3 TB/s); this form is
// ~13 ops (v_rcp + v_exp are quarter-rate HW instructions).
__device__ __forceinline__ float erf_fast(float z) {
  float az = fabsf(z);
  float t = 1.f / (1.f + 0.3275911f * az);
  float p = t * (0.254829592f + t * (-0.284496736f + t * (1.421413741f +
            t * (-1.453152027f + t * 1.061405429f))));
  float r = 1.f - p * __expf(-az * az);
  return copysignf(r, z);
}
// exact-erf GeLU to match paddle.nn.GELU(approximate=False)
__device__ __forceinline__ float gelu_f(float x) {
  return 0.5f * x * (1.f + erf_fast(x * 0.70710678118654752440f));
}
__device__ __forceinline__ float gelu_grad_f(float x) {
  const float kInvSqrt2 = 0.70710678118654752440f;
  const float kInvSqrt2Pi = 0.3989422804014327f;
  float cdf = 0.5f * (1.f + erf_fast(x * kInvSqrt2));
  float pdf = kInvSqrt2Pi * __expf(-0.5f * x * x);
  return cdf + x * pdf;
}
__device__ __forceinline__ float silu_f(float x) {
  return x / (1.f + __expf(-x));
}
__device__ __forceinline__ float silu_grad_f(float x) {
  float sig = 1.f / (1.f + __expf(-x));
  return sig * (1.f + x * (1.f - sig));
}

template <int DT, bool FWD, int W>
__global__ void bias_gelu_kernel(const void* __restrict__ a, const void* __restrict__ x,
                                 const void* __restrict__ bias, void* __restrict__ out,
                                 int64_t n, int64_t d) {
  // FWD: a == x (unused), out = gelu(x+bias).  BWD: a = dy, out = dx.
  // W=16: two 16B loads in flight per stream -- streaming-friendly MLP.
  int64_t total = n * d;
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * W;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * W;
  // bias column tracked incrementally -- no 64-bit modulo in the loop
  // (d divides both W-chunks and wraps cleanly: launcher guarantees d % W == 0
  // alignment of every chunk start)
  int64_t c = i0 % d, cstep = stride % d;
  for (int64_t i = i0; i < total; i += stride, c = (c + cstep >= d ? c + cstep - d : c + cstep)) {
    float xf[W], bf[W], af[W];
#pragma unroll
    for (int v8 = 0; v8 < W / 8; ++v8) {
      LS8<DT>::load8(x, i + 8 * v8, xf + 8 * v8);
      if (bias) LS8<DT>::load8(bias, c + 8 * v8, bf + 8 * v8);
      if (!FWD) LS8<DT>::load8(a, i + 8 * v8, af + 8 * v8);
    }
#pragma unroll
    for (int k = 0; k < W; ++k) {
      float v = xf[k] + (bias ? bf[k] : 0.f);
      xf[k] = FWD ? gelu_f(v) : af[k] * gelu_grad_f(v);
    }
#pragma unroll
    for (int v8 = 0; v8 < W / 8; ++v8)
      LS8<DT>::store8(out, i + 8 * v8, xf + 8 * v8);
  }
}

template <int DT>
__global__ void swiglu_fwd_kernel(const void* __restrict__ x, void* __restrict__ y,
                                  int64_t n, int64_t d) {
  int64_t total = n * d;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8; i < total;
       i += (int64_t)gridDim.x * blockDim.x * 8) {
    int64_t row = i / d, col = i % d;
    float g[8], u[8];
    LS8<DT>::load8(x, row * 2 * d + col, g);
    LS8<DT>::load8(x, row * 2 * d + d + col, u);
#pragma unroll
    for (int k = 0; k < 8; ++k) g[k] = silu_f(g[k]) * u[k];
    LS8<DT>::store8(y, i, g);
  }
}

template <int DT>
__global__ void swiglu_bwd_kernel(const void* __restrict__ dy, const void* __restrict__ x,
                                  void* __restrict__ dx, int64_t n, int64_t d) {
  int64_t total = n * d;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8; i < total;
       i += (int64_t)gridDim.x * blockDim.x * 8) {
    int64_t row = i / d, col = i % d;
    float g[8], u[8], dyf[8], dg[8];
    LS8<DT>::load8(x, row * 2 * d + col, g);
    LS8<DT>::load8(x, row * 2 * d + d + col, u);
    LS8<DT>::load8(dy, i, dyf);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      dg[k] = dyf[k] * u[k] * silu_grad_f(g[k]);
      u[k] = dyf[k] * silu_f(g[k]);  // du
    }
    LS8<DT>::store8(dx, row * 2 * d + col, dg);
    LS8<DT>::store8(dx, row * 2 * d + d + col, u);
  }
}

// rope rotate-half (neox style): for pair (x1=x[..,j], x2=x[..,j+dh/2]):
//   y1 = x1*cos[j] - x2*sin[j];  y2 = x2*cos[j] + x1*sin[j]
// conj (backward): sin -> -sin.
template <int DT>
__global__ void rope_kernel(const void* __restrict__ x, const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t, void* __restrict__ y,
                            int64_t b, int64_t sl, int64_t h, int64_t dh,
                            int64_t pos_offset, float sin_sign) {
  // one (b, s, h) row per 64-lane wave; dh/2 pairs processed 4-at-a-time
  int64_t rows = b * sl * h;
  int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int lane = threadIdx.x & 63;
  int64_t nwaves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  int64_t half = dh / 2;
  for (int64_t r = wid; r < rows; r += nwaves) {
    int64_t srow = (r / h) % sl + pos_offset;
    const int64_t base = r * dh;
    for (int64_t j = lane * 4; j < half; j += 64 * 4) {
      float x1[4], x2[4];
      // load 4 elements from each half (4*bf16 = 8B)
      if (DT == kBF16) {
        shortx4 v1 = *reinterpret_cast<const shortx4*>((const short*)x + base + j);
        shortx4 v2 = *reinterpret_cast<const shortx4*>((const short*)x + base + half + j);
#pragma unroll
        for (int k = 0; k < 4; ++k) { x1[k] = bf2f(v1[k]); x2[k] = bf2f(v2[k]); }
      } else {
        const float* xf = (const float*)x;
#pragma unroll
        for (int k = 0; k < 4; ++k) { x1[k] = xf[base + j + k]; x2[k] = xf[base + half + j + k]; }
      }
      const float4 c = *reinterpret_cast<const float4*>(&cos_t[srow * half + j]);
      const float4 sn = *reinterpret_cast<const float4*>(&sin_t[srow * half + j]);
      float cc[4] = {c.x, c.y, c.z, c.w};
      float ss[4] = {sn.x * sin_sign, sn.y * sin_sign, sn.z * sin_sign, sn.w * sin_sign};
      float y1[4], y2[4];
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        y1[k] = x1[k] * cc[k] - x2[k] * ss[k];
        y2[k] = x2[k] * cc[k] + x1[k] * ss[k];
      }
      if (DT == kBF16) {
        shortx4 o1, o2;
#pragma unroll
        for (int k = 0; k < 4; ++k) { o1[k] = f2bf(y1[k]); o2[k] = f2bf(y2[k]); }
        *reinterpret_cast<shortx4*>((short*)y + base + j) = o1;
        *reinterpret_cast<shortx4*>((short*)y + base + half + j) = o2;
      } else {
        float* yf = (float*)y;
#pragma unroll
        for (int k = 0; k < 4; ++k) { yf[base + j + k] = y1[k]; yf[base + half + j + k] = y2[k]; }
      }
    }
  }
}

// column-sum [n, d] -> fp32 [d].  8-wide column strips per thread (16 B
// loads) instead of scalar b16 loads; 2-D grid tiles rows.
template <int DT>
__global__ void colsum_kernel(const void* __restrict__ x, float* __restrict__ out,
                              int64_t n, int64_t d) {
  int64_t c0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (c0 >= d) return;
  int64_t rows_per = (n + gridDim.y - 1) / gridDim.y;
  int64_t r0 = (int64_t)blockIdx.y * rows_per, r1 = min(n, r0 + rows_per);
  float acc[8] = {0.f};
  if (c0 + 8 <= d) {
    for (int64_t r = r0; r < r1; ++r) {
      float v[8];
      LS8<DT>::load8(x, r * d + c0, v);
#pragma unroll
      for (int k = 0; k < 8; ++k) acc[k] += v[k];
    }
  } else {
    for (int64_t r = r0; r < r1; ++r)
      for (int64_t c = c0; c < d; ++c)
        acc[c - c0] += (DT == kBF16) ? bf2f(((const short*)x)[r * d + c])
                                     : ((const float*)x)[r * d + c];
  }
#pragma unroll
  for (int k = 0; k < 8; ++k)
    if (c0 + k < d) atomicAdd(&out[c0 + k], acc[k]);
}

// Philox-free dropout: xorshift per-element hash of (seed, offset+idx).
// Deterministic given (seed, offset) -- enough for recompute parity.
__device__ __forceinline__ float rand_uniform(uint64_t seed, uint64_t idx) {
  uint64_t z = seed + idx * 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  z = z ^ (z >> 31);
  return (float)(z >> 40) * (1.f / 16777216.f);
}

template <int DT>
__global__ void dropout_add_fwd_kernel(const void* __restrict__ x, const void* __restrict__ res,
                                       void* __restrict__ y, uint8_t* __restrict__ mask,
                                       int64_t numel, float p, uint64_t seed, uint64_t offset) {
  const float scale_ = 1.f / (1.f - p);
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8; i < numel;
       i += (int64_t)gridDim.x * blockDim.x * 8) {
    float xf[8], rf[8];
    LS8<DT>::load8(x, i, xf);
    if (res) LS8<DT>::load8(res, i, rf);
    uchar2 mk[4];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      bool keep = p == 0.f || rand_uniform(seed, offset + i + k) >= p;
      ((uint8_t*)mk)[k] = keep;
      xf[k] = (keep ? xf[k] * scale_ : 0.f) + (res ? rf[k] : 0.f);
    }
    if (mask) *reinterpret_cast<uint2*>(mask + i) = *reinterpret_cast<uint2*>(mk);
    LS8<DT>::store8(y, i, xf);
  }
}

template <int DT>
__global__ void dropout_add_bwd_kernel(const void* __restrict__ dy, const uint8_t* __restrict__ mask,
                                       void* __restrict__ dx, int64_t numel, float p) {
  const float scale_ = 1.f / (1.f - p);
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8; i < numel;
       i += (int64_t)gridDim.x * blockDim.x * 8) {
    float f[8];
    LS8<DT>::load8(dy, i, f);
#pragma unroll
    for (int k = 0; k < 8; ++k) f[k] = mask[i + k] ? f[k] * scale_ : 0.f;
    LS8<DT>::store8(dx, i, f);
  }
}

// ---- host launchers -------------------------------------------------------
#define EDT(dtype, KERNEL_CALL_BF16, KERNEL_CALL_F32) \
  if (dtype == kBF16) { KERNEL_CALL_BF16; } else { KERNEL_CALL_F32; }

static dim3 egrid(int64_t numel) {
  return dim3((unsigned)elementwise_grid(cdiv((int)hmin<int64_t>(numel / 8 + 1, 1 << 30), 256)));
}

void bias_gelu_fwd(const void* x, const void* bias, void* y, int64_t n,
                   int64_t d, int dtype, hipStream_t s) {
  if (d % 16 == 0) {
    dim3 g = egrid(n * d / 2);
    EDT(dtype,
        hipLaunchKernelGGL((bias_gelu_kernel<kBF16, true, 16>), g, dim3(256), 0, s, x, x, bias, y, n, d),
        hipLaunchKernelGGL((bias_gelu_kernel<kF32, true, 16>), g, dim3(256), 0, s, x, x, bias, y, n, d));
  } else {
    dim3 g = egrid(n * d);
    EDT(dtype,
        hipLaunchKernelGGL((bias_gelu_kernel<kBF16, true, 8>), g, dim3(256), 0, s, x, x, bias, y, n, d),
        hipLaunchKernelGGL((bias_gelu_kernel<kF32, true, 8>), g, dim3(256), 0, s, x, x, bias, y, n, d));
  }
}

void bias_gelu_bwd(const void* dy, const void* x, const void* bias, void* dx,
                   int64_t n, int64_t d, int dtype, hipStream_t s) {
  if (d % 16 == 0) {
    dim3 g = egrid(n * d / 2);
    EDT(dtype,
        hipLaunchKernelGGL((bias_gelu_kernel<kBF16, false, 16>), g, dim3(256), 0, s, dy, x, bias, dx, n, d),
        hipLaunchKernelGGL((bias_gelu_kernel<kF32, false, 16>), g, dim3(256), 0, s, dy, x, bias, dx, n, d));
  } else {
    dim3 g = egrid(n * d);
    EDT(dtype,
        hipLaunchKernelGGL((bias_gelu_kernel<kBF16, false, 8>), g, dim3(256), 0, s, dy, x, bias, dx, n, d),
        hipLaunchKernelGGL((bias_gelu_kernel<kF32, false, 8>), g, dim3(256), 0, s, dy, x, bias, dx, n, d));
  }
}

void swiglu_fwd(const void* x, void* y, int64_t n, int64_t d, int dtype, hipStream_t s) {
  dim3 g = egrid(n * d);
  EDT(dtype,
      hipLaunchKernelGGL((swiglu_fwd_kernel<kBF16>), g, dim3(256), 0, s, x, y, n, d),
      hipLaunchKernelGGL((swiglu_fwd_kernel<kF32>), g, dim3(256), 0, s, x, y, n, d));
}

void swiglu_bwd(const void* dy, const void* x, void* dx, int64_t n, int64_t d,
                int dtype, hipStream_t s) {
  dim3 g = egrid(n * d);
  EDT(dtype,
      hipLaunchKernelGGL((swiglu_bwd_kernel<kBF16>), g, dim3(256), 0, s, dy, x, dx, n, d),
      hipLaunchKernelGGL((swiglu_bwd_kernel<kF32>), g, dim3(256), 0, s, dy, x, dx, n, d));
}

void rope_fwd(const void* x, const float* cos_t, const float* sin_t, void* y,
              int64_t b, int64_t sl, int64_t h, int64_t dh, int64_t pos_offset,
              bool conj, int dtype, hipStream_t s) {
  int64_t rows = b * sl * h;
  dim3 g((unsigned)hmin<int64_t>(cdiv((int)hmin<int64_t>(rows, 1 << 24), 4), 2048));
  float ssign = conj ? -1.f : 1.f;
  EDT(dtype,
      hipLaunchKernelGGL((rope_kernel<kBF16>), g, dim3(256), 0, s, x, cos_t, sin_t, y, b, sl, h, dh, pos_offset, ssign),
      hipLaunchKernelGGL((rope_kernel<kF32>), g, dim3(256), 0, s, x, cos_t, sin_t, y, b, sl, h, dh, pos_offset, ssign));
}

void colsum(const void* x, float* out, int64_t n, int64_t d, int dtype, hipStream_t s) {
  int xblocks = cdiv((int)d, 256 * 8);
  // keep >=1024 workgroups in flight (256 CUs / 8 XCDs want oversubscription)
  int chunks = (int)hmin<int64_t>(hmax<int64_t>(1, n / 32),
                                  hmax<int64_t>(1, 2048 / xblocks));
  dim3 grid((unsigned)xblocks, chunks);
  EDT(dtype,
      hipLaunchKernelGGL((colsum_kernel<kBF16>), grid, dim3(256), 0, s, x, out, n, d),
      hipLaunchKernelGGL((colsum_kernel<kF32>), grid, dim3(256), 0, s, x, out, n, d));
}

void dropout_add_fwd(const void* x, const void* residual, void* y,
                     uint8_t* mask, int64_t numel, float p, uint64_t seed,
                     uint64_t offset, int dtype, hipStream_t s) {
  dim3 g = egrid(numel);
  EDT(dtype,
      hipLaunchKernelGGL((dropout_add_fwd_kernel<kBF16>), g, dim3(256), 0, s, x, residual, y, mask, numel, p, seed, offset),
      hipLaunchKernelGGL((dropout_add_fwd_kernel<kF32>), g, dim3(256), 0, s, x, residual, y, mask, numel, p, seed, offset));
}

void dropout_add_bwd(const void* dy, const uint8_t* mask, void* dx,
                     int64_t numel, float p, int dtype, hipStream_t s) {
  dim3 g = egrid(numel);
  EDT(dtype,
      hipLaunchKernelGGL((dropout_add_bwd_kernel<kBF16>), g, dim3(256), 0, s, dy, mask, dx, numel, p),
      hipLaunchKernelGGL((dropout_add_bwd_kernel<kF32>), g, dim3(256), 0, s, dy, mask, dx, numel, p));
}

}  // namespace pa

namespace pa {
// ---------------------------------------------------------------------------
// fused |x| max: single read pass (x.abs().amax() materializes |x| and
// re-reads it -- 3x the traffic; this is the fp8 quantize-scale producer)
// ---------------------------------------------------------------------------
template <int DT>
__global__ void amax_abs_kernel(const void* __restrict__ x, float* __restrict__ out,
                                int64_t n) {
  float m = 0.f;
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = i0; i + 8 <= n; i += stride) {
    float v[8];
    LS8<DT>::load8(x, i, v);
#pragma unroll
    for (int k = 0; k < 8; ++k) m = fmaxf(m, fabsf(v[k]));
  }
  // scalar tail (last partial group)
  int64_t tail = n & ~7ll;
  if (i0 == 0)
    for (int64_t i = tail; i < n; ++i)
      m = fmaxf(m, fabsf(DT == kBF16
          ? bf2f(reinterpret_cast<const short*>(x)[i])
          : reinterpret_cast<const float*>(x)[i]));
#pragma unroll
  for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  if ((threadIdx.x & 63) == 0)
    atomicMax(reinterpret_cast<int*>(out), __float_as_int(m));  // m >= 0
}

void amax_abs(const void* x, float* out, int64_t n, int dtype, hipStream_t s) {
  int64_t waves = (n + 8 * 256 - 1) / (8 * 256);
  unsigned g = (unsigned)hmin<int64_t>(hmax<int64_t>(waves, 1), 4096);
  EDT(dtype,
      hipLaunchKernelGGL((amax_abs_kernel<kBF16>), dim3(g), dim3(256), 0, s, x, out, n),
      hipLaunchKernelGGL((amax_abs_kernel<kF32>), dim3(g), dim3(256), 0, s, x, out, n));
}
}  // namespace pa
