// Common device helpers for paddle_amd gfx950 (CDNA4) kernels.
//
// MI355X-first conventions used across csrc/:
//   * wavefront = 64 lanes; block sizes are multiples of 64
//   * bf16 data is loaded/stored vectorized (>=8 bytes per lane)
//   * fp32 accumulation everywhere; MFMA 16x16x32 bf16 tiles for
//     matmul-shaped work
//   * grids are sized >> 256 workgroups to fill 8 XCDs
//
// Reference parity: this layer replaces paddle/phi/kernels/funcs/
// (elementwise_base.h, reduce_function.h) and kernels/primitive/ -- the
// block-level primitives are re-derived for 64-wide waves, not ported.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,    \
             __LINE__);                                                     \
    }                                                                       \
  } while (0)

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(2))) float floatx2;
typedef __attribute__((ext_vector_type(8))) short shortx8;   // 8 bf16 = 16B
typedef __attribute__((ext_vector_type(4))) short shortx4;   // 8B
typedef __attribute__((ext_vector_type(4))) int intx4;

__device__ __forceinline__ float bf2f(short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

// RNE via the hardware convert (v_cvt); bit-math emulation costs ~6 VALU ops
__device__ __forceinline__ short f2bf(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

// float -> bf16 for a vector of 8
__device__ __forceinline__ shortx8 f8_to_bf8(const float *f) {
  shortx8 r;
#pragma unroll
  for (int i = 0; i < 8; ++i) r[i] = f2bf(f[i]);
  return r;
}

// ---------------------------------------------------------------------------
// Wave + block reductions (64-lane; replaces KPS compute_primitives.h)
// ---------------------------------------------------------------------------
template <typename Op>
__device__ __forceinline__ float wave_reduce(float v, Op op) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, WAVE));
  return v;
}

struct SumOp { __device__ float operator()(float a, float b) const { return a + b; } };
struct MaxOp { __device__ float operator()(float a, float b) const { return fmaxf(a, b); } };

// Block-wide reduce for blockDim.x == 256 (4 waves). `lds` needs >= 4 floats.
template <typename Op>
__device__ __forceinline__ float block_reduce_256(float v, Op op, float *lds, float init) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x >> 6;
  v = wave_reduce(v, op);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float r = init;
#pragma unroll
  for (int i = 0; i < 4; ++i) r = op(r, lds[i]);
  return r;
}

// Reduce within a 16-lane group (xor over 1,2,4,8): used for MFMA row stats
// where the 16 lanes holding one C-tile row-group must agree.
template <typename Op>
__device__ __forceinline__ float group16_reduce(float v, Op op) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, WAVE));
  return v;
}

// ---------------------------------------------------------------------------
// MFMA wrappers (gfx950): D(f32 16x16) = A(bf16 16x32) * B(bf16 32x16) + C
// Fragment layouts (verified per guide / learn_hip m89):
//   A: lane l holds A[row = l%16][k = 8*(l/16) + 0..7]
//   B: lane l holds B[k = 8*(l/16) + 0..7][col = l%16]
//   C/D: lane l reg r holds D[row = 4*(l/16) + r][col = l%16]
// ---------------------------------------------------------------------------
__device__ __forceinline__ floatx4 mfma_bf16(shortx8 a, shortx8 b, floatx4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// XOR swizzle for LDS rows that are read column-sliced (guide §6 G4):
// spread a 32-way bank conflict across 8 distinct 16B slots.
__device__ __forceinline__ unsigned swz(unsigned row, unsigned byte_in_row) {
  return byte_in_row ^ ((row & 7u) << 4);
}

// ceil-div
__host__ __device__ __forceinline__ int cdiv(int a, int b) { return (a + b - 1) / b; }

// host-side min/max (device min/max from HIP headers aren't host-callable)
template <typename T> inline T hmin(T a, T b) { return a < b ? a : b; }
template <typename T> inline T hmax(T a, T b) { return a > b ? a : b; }

// Grid sizing for memory-bound grid-stride kernels: cap at ~8 blocks/CU.
inline int elementwise_grid(long long n_blocks_needed) {
  long long cap = 256LL * 8;  // 256 CUs x 8 blocks
  return (int)(n_blocks_needed < cap ? n_blocks_needed : cap);
}

// ---------------------------------------------------------------------------
// Philox4x32-10 counter RNG (reference funcs/dropout_impl.cu.h:129 keeps
// (seed, offset) pairs so recompute replays the same mask).  One call
// yields 4 uint32 lanes for 4 consecutive elements.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(4))) unsigned int uintx4;

__device__ __forceinline__ uintx4 philox10(unsigned long long seed,
                                           unsigned long long ctr) {
  unsigned key0 = (unsigned)seed, key1 = (unsigned)(seed >> 32);
  unsigned c0 = (unsigned)ctr, c1 = (unsigned)(ctr >> 32), c2 = 0, c3 = 0;
#pragma unroll
  for (int i = 0; i < 10; ++i) {
    unsigned hi0 = __umulhi(0xD2511F53u, c0), lo0 = 0xD2511F53u * c0;
    unsigned hi1 = __umulhi(0xCD9E8D57u, c2), lo1 = 0xCD9E8D57u * c2;
    c0 = hi1 ^ c1 ^ key0; c1 = lo1;
    c2 = hi0 ^ c3 ^ key1; c3 = lo0;
    key0 += 0x9E3779B9u; key1 += 0xBB67AE85u;
  }
  uintx4 r = {c0, c1, c2, c3};
  return r;
}

// keep-decision for attention dropout at element (q_abs, kv_abs) of head
// slot bh: the counter is the global element index / 4 (4 consecutive kv
// share one philox call); threshold compare in 24-bit space.
__device__ __forceinline__ bool fa_keep(unsigned long long seed,
                                        unsigned long long offset,
                                        long long bh, int sq_, long long skv_,
                                        int q_abs, long long kv_abs,
                                        unsigned thr24) {
  long long elem = ((bh * sq_ + q_abs) * skv_ + kv_abs);
  uintx4 r = philox10(seed, offset + (unsigned long long)(elem >> 2));
  unsigned v = r[elem & 3];
  return (v >> 8) >= thr24;
}
