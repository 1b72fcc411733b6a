// fp8 (OCP e4m3) MX-scaled MFMA GEMM for gfx950: C = A x B^T, fp32
// accumulate, bf16 out, per-tensor float scales applied in the epilogue.
//
// Reference role: paddle/phi/kernels/fusion/cutlass/ fp8 GEMM paths --
// re-derived natively on the gfx950-only block-scaled MX pipeline
// (mfma_scale_f32_16x16x128_f8f6f4, the ONLY large-K fp8 MFMA; layout
// verified on hardware by tools/probe_fp8mx.py: A lane frag = row l%16,
// k-bytes 32*(l/16)..+31; C same as 16x16x32; e8m0 scale bytes, 127=1.0).
// Runs at ~2x the bf16 MFMA rate; block scales are set to 1.0 and the
// per-tensor dequant scale (sa*sb) multiplies the accumulator once.
//
// Structure: the bf16 8-phase-derived 4-phase counted-vmcnt pipeline from
// gemm.hip, reused byte-for-byte -- an fp8 [256][128] tile has the same
// byte geometry as bf16 [256][64] (128 B rows, 16 B-granule XOR swizzle).
#include "common.h"
#include "api.h"

namespace pa {

typedef __attribute__((ext_vector_type(8))) int intx8;

__device__ __forceinline__ floatx4 mfma_fp8mx(intx8 a, intx8 b, floatx4 c) {
  // fmt 0/0 = e4m3/e4m3; scale words 0x7F7F7F7F = 1.0 per 32-elem block
  return __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      a, b, c, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
}

// byte offset in a [256][128B] LDS tile with the 16B-granule row XOR
__device__ __forceinline__ unsigned f8_swz(unsigned row, unsigned bcol) {
  return row * 128 + (bcol ^ ((row & 7u) << 4));
}

template <bool FAST>
__launch_bounds__(512, 2)
__global__ void gemm_fp8_kernel(const unsigned char* __restrict__ ag,
                                const unsigned char* __restrict__ bg,
                                short* __restrict__ cg,
                                const short* __restrict__ biasg,
                                float sab, int M, int N, int K,
                                long long lda, long long ldb, long long ldc,
                                int skip_interior, int has_bias,
                                long long a_bs = 0, long long b_bs = 0,
                                long long c_bs = 0, long long bias_bs = 0) {
  // batched (grouped-expert) mode: blockIdx.z selects the problem
  ag += (long long)blockIdx.z * a_bs;
  bg += (long long)blockIdx.z * b_bs;
  cg += (long long)blockIdx.z * c_bs;
  if (has_bias) biasg += (long long)blockIdx.z * bias_bs;
  constexpr int BM = 256, BN = 256, BK = 128;
  __shared__ unsigned char a_lds[2][BM * BK];
  __shared__ unsigned char b_lds[2][BN * BK];

  const int nwg = gridDim.x * gridDim.y;
  int orig = blockIdx.y * gridDim.x + blockIdx.x;
  {
    const int nx = 8;
    int q = nwg / nx, rr = nwg % nx;
    int xcd = orig % nx, pos = orig / nx;
    orig = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + pos;
  }
  const int mt = FAST ? (M / BM) : ((M + BM - 1) / BM);
  const int nt = FAST ? (N / BN) : ((N + BN - 1) / BN);
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
  if (!FAST && skip_interior &&
      row0 + BM <= M && col0 + BN <= N && (K % BK) == 0)
    return;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;
  const int wc = wid & 3;
  const int l16 = lane & 15;
  const int lg = lane >> 4;

  // DMA staging map: 4 rounds x (512 thr x 16 B); round r covers 64 rows
  int s_r[4], s_cp[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    int fb = it * 8192 + tid * 16;
    int r = fb / BK, c = fb % BK;
    s_r[it] = r;
    s_cp[it] = c ^ ((r & 7) << 4);
  }
  const int nkt = (K + BK - 1) / BK;

  auto stage_a_round = [&](int buf, int kt, int it) {
    const long long k0 = (long long)kt * BK;
    const unsigned char* src = ag + (long long)(row0 + s_r[it]) * lda + k0 + s_cp[it];
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)&a_lds[buf][it * 8192 + tid * 16],
        16, 0, 0);
  };
  auto stage_a_dma = [&](int buf, int kt) {
#pragma unroll
    for (int it = 0; it < 4; ++it) stage_a_round(buf, kt, it);
  };
  auto stage_b_dma = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const unsigned char* src = bg + (long long)(col0 + s_r[it]) * ldb + k0 + s_cp[it];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)&b_lds[buf][it * 8192 + tid * 16],
          16, 0, 0);
    }
  };
  auto stage_guarded = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
    for (int fb = tid * 16; fb < BM * BK; fb += 512 * 16) {
      int r = fb / BK, c = fb % BK;
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        // A then B, scalar-guarded 16B chunks
        const unsigned char* src = half ? bg : ag;
        long long ld = half ? ldb : lda;
        int lim = half ? N : M;
        int base0 = half ? col0 : row0;
        unsigned char v[16];
        if (base0 + r < lim && k0 + c + 15 < K) {
          *reinterpret_cast<intx4*>(v) = *reinterpret_cast<const intx4*>(
              src + (long long)(base0 + r) * ld + k0 + c);
        } else {
#pragma unroll
          for (int j = 0; j < 16; ++j)
            v[j] = (base0 + r < lim && k0 + c + j < K)
                       ? src[(long long)(base0 + r) * ld + k0 + c + j] : 0;
        }
        unsigned char* dst = half ? b_lds[buf] : a_lds[buf];
        *reinterpret_cast<intx4*>(&dst[f8_swz(r, c)]) = *reinterpret_cast<intx4*>(v);
      }
    }
  };

  floatx4 acc[8][4];
#pragma unroll
  for (int m = 0; m < 8; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  // fragment regs: 32 B/lane per frag (two b128 reads at lg*32, lg*32+16)
  intx8 afA[4], afB[4];   // A(m0) / A(m1), 4 mfrags each
  intx8 bfA[2], bfB[2];   // B(n0) / B(n1), 2 nfrags each

  auto read_a_half = [&](int buf, int mh, intx8* dst) {
#pragma unroll
    for (int m = 0; m < 4; ++m) {
      unsigned row = wr * 128 + (mh * 4 + m) * 16 + l16;
      intx4* d = reinterpret_cast<intx4*>(&dst[m]);
      d[0] = *reinterpret_cast<const intx4*>(&a_lds[buf][f8_swz(row, lg * 32)]);
      d[1] = *reinterpret_cast<const intx4*>(&a_lds[buf][f8_swz(row, lg * 32 + 16)]);
    }
  };
  auto read_b_half = [&](int buf, int nh, intx8* dst) {
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      unsigned row = wc * 64 + (nh * 2 + n) * 16 + l16;
      intx4* d = reinterpret_cast<intx4*>(&dst[n]);
      d[0] = *reinterpret_cast<const intx4*>(&b_lds[buf][f8_swz(row, lg * 32)]);
      d[1] = *reinterpret_cast<const intx4*>(&b_lds[buf][f8_swz(row, lg * 32 + 16)]);
    }
  };
  auto mfma_quadrant = [&](intx8* af, intx8* bf, int mh, int nh) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 2; ++n)
        acc[mh * 4 + m][nh * 2 + n] =
            mfma_fp8mx(af[m], bf[n], acc[mh * 4 + m][nh * 2 + n]);
    __builtin_amdgcn_s_setprio(0);
  };

  if constexpr (FAST) {
    // simple 2-barrier DMA pipeline (the bf16 kernel's deeper 4-phase
    // schedule spilled here: fp8 fragments are 2x the registers)
    stage_a_dma(0, 0);
    stage_b_dma(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    for (int kt = 0; kt < nkt; ++kt) {
      const int cur = kt & 1;
      if (kt + 1 < nkt) { stage_a_dma(cur ^ 1, kt + 1); stage_b_dma(cur ^ 1, kt + 1); }
      read_a_half(cur, 0, afA);
      read_b_half(cur, 0, bfA);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(afA, bfA, 0, 0);
      read_b_half(cur, 1, bfB);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(afA, bfB, 0, 1);
      read_a_half(cur, 1, afB);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(afB, bfB, 1, 1);
      mfma_quadrant(afB, bfA, 1, 0);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    }
  } else {
    stage_guarded(0, 0);
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __syncthreads();
    for (int kt = 0; kt < nkt; ++kt) {
      const int cur = kt & 1;
      if (kt + 1 < nkt) stage_guarded(cur ^ 1, kt + 1);
      read_a_half(cur, 0, afA);
      read_b_half(cur, 0, bfA);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(afA, bfA, 0, 0);
      read_b_half(cur, 1, bfB);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(afA, bfB, 0, 1);
      read_a_half(cur, 1, afB);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      mfma_quadrant(afB, bfB, 1, 1);
      mfma_quadrant(afB, bfA, 1, 0);
      asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
      __syncthreads();
    }
  }

  // epilogue: per-tensor dequant scale + optional bias, bf16 store
  float bias_v[4];
  if (has_bias) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int col = col0 + wc * 64 + nf * 16 + l16;
      bias_v[nf] = (col < N) ? bf2f(biasg[col]) : 0.f;
    }
  }
#pragma unroll
  for (int m = 0; m < 8; ++m) {
    int row = row0 + wr * 128 + m * 16 + (lg * 4);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (row + r >= M) continue;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int col = col0 + wc * 64 + nf * 16 + l16;
        if (col >= N) continue;
        float v = acc[m][nf][r] * sab;
        if (has_bias) v += bias_v[nf];
        cg[(long long)(row + r) * ldc + col] = f2bf(v);
      }
    }
  }
}

void gemm_fp8_nt_batched(const void* a, const void* bt, void* c,
                         const void* bias, float scale_ab, int64_t batch,
                         int64_t m, int64_t n, int64_t k, int64_t lda,
                         int64_t ldb, int64_t ldc, int64_t a_bs, int64_t b_bs,
                         int64_t c_bs, hipStream_t s, int64_t bias_bs) {
  const int mi = (int)(m / 256), ni = (int)(n / 256);
  const int gm = (int)((m + 255) / 256), gn = (int)((n + 255) / 256);
  const bool k_ok = (k % 128 == 0);
  const bool has_fast = k_ok && mi > 0 && ni > 0;
  const bool has_edge = !has_fast || mi < gm || ni < gn;
  dim3 blk(512);
  if (has_fast)
    hipLaunchKernelGGL((gemm_fp8_kernel<true>),
                       dim3((unsigned)mi, (unsigned)ni, (unsigned)batch),
                       blk, 0, s, (const unsigned char*)a, (const unsigned char*)bt,
                       (short*)c, (const short*)bias, scale_ab, (int)m, (int)n,
                       (int)k, lda, ldb, ldc, 0, bias != nullptr,
                       a_bs, b_bs, c_bs, bias_bs);
  if (has_edge)
    hipLaunchKernelGGL((gemm_fp8_kernel<false>),
                       dim3((unsigned)gm, (unsigned)gn, (unsigned)batch),
                       blk, 0, s, (const unsigned char*)a, (const unsigned char*)bt,
                       (short*)c, (const short*)bias, scale_ab, (int)m, (int)n,
                       (int)k, lda, ldb, ldc, has_fast ? 1 : 0, bias != nullptr,
                       a_bs, b_bs, c_bs, bias_bs);
}

void gemm_fp8_nt(const void* a, const void* bt, void* c, const void* bias,
                 float scale_ab, int64_t m, int64_t n, int64_t k, int64_t lda,
                 int64_t ldb, int64_t ldc, hipStream_t s) {
  gemm_fp8_nt_batched(a, bt, c, bias, scale_ab, 1, m, n, k, lda, ldb, ldc,
                      0, 0, 0, s, 0);
}

// bf16 -> e4m3 cast with a uniform scale, packed x8 (v_cvt_pk_fp8_f32);
// one read + one half-size write, no fp32 materialization
__global__ void quant_fp8_kernel(const short* __restrict__ x,
                                 unsigned char* __restrict__ out,
                                 float scale, long long n8) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n8; i += stride) {
    shortx8 v = *reinterpret_cast<const shortx8*>(x + i * 8);
    int2 packed;
    int p0 = 0, p1 = 0;
    p0 = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f(v[0]) * scale, bf2f(v[1]) * scale, p0, false);
    p0 = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f(v[2]) * scale, bf2f(v[3]) * scale, p0, true);
    p1 = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f(v[4]) * scale, bf2f(v[5]) * scale, p1, false);
    p1 = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f(v[6]) * scale, bf2f(v[7]) * scale, p1, true);
    packed.x = p0;
    packed.y = p1;
    *reinterpret_cast<int2*>(out + i * 8) = packed;
  }
}

void quant_fp8(const void* x, void* out, float scale, int64_t numel,
               hipStream_t s) {
  long long n8 = numel / 8;
  dim3 g((unsigned)hmin<long long>((n8 + 255) / 256, 2048));
  hipLaunchKernelGGL(quant_fp8_kernel, g, dim3(256), 0, s, (const short*)x,
                     (unsigned char*)out, scale, n8);
}

}  // namespace pa
