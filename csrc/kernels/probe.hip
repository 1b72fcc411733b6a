// MFMA layout probe: single-wave 16x16x32 bf16 tile, used by the GPU test
// suite to verify the fragment layout assumptions in flash_attn.hip /
// gemm kernels against torch.matmul (asymmetric-input check, guide G9).
// No reference counterpart: this exists because CDNA4 fragment layouts
// are hardware facts the kernels above depend on (SURVEY §2.2 notes).
#include "common.h"
#include "api.h"

namespace pa {

__global__ void mfma_probe_kernel(const short* __restrict__ a, const short* __restrict__ b,
                                  float* __restrict__ c) {
  // a: [16][32] row-major bf16 ; b_t: [16][32] row-major (i.e. B^T, B is [32][16])
  // c: [16][16] row-major fp32 ; launched with 64 threads (one wave)
  int lane = threadIdx.x & 63;
  int l16 = lane & 15, lg = lane >> 4;
  shortx8 af = *reinterpret_cast<const shortx8*>(a + l16 * 32 + lg * 8);
  shortx8 bf = *reinterpret_cast<const shortx8*>(b + l16 * 32 + lg * 8);
  floatx4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = mfma_bf16(af, bf, acc);
#pragma unroll
  for (int r = 0; r < 4; ++r) c[(lg * 4 + r) * 16 + l16] = acc[r];
}

void mfma_probe(const void* a, const void* bt, float* c, hipStream_t s);
void mfma_probe(const void* a, const void* bt, float* c, hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, s,
                     (const short*)a, (const short*)bt, c);
}

}  // namespace pa

namespace pa {
typedef __attribute__((ext_vector_type(16))) float floatx16;

__global__ void mfma_probe32_kernel(const short* __restrict__ a, const short* __restrict__ b,
                                    float* __restrict__ c) {
  // a: [32][16] row-major bf16 ; b_t: [32][16] row-major (B^T, B is [16][32])
  // c: [32][32] row-major fp32 ; one wave
  int lane = threadIdx.x & 63;
  int l32 = lane & 31, hi = lane >> 5;
  shortx8 af = *reinterpret_cast<const shortx8*>(a + l32 * 16 + hi * 8);
  shortx8 bf = *reinterpret_cast<const shortx8*>(b + l32 * 16 + hi * 8);
  floatx16 acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    c[((r & 3) + 8 * (r >> 2) + 4 * hi) * 32 + l32] = acc[r];
}

void mfma_probe32(const void* a, const void* bt, float* c, hipStream_t s);
void mfma_probe32(const void* a, const void* bt, float* c, hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe32_kernel, dim3(1), dim3(64), 0, s,
                     (const short*)a, (const short*)bt, c);
}
}  // namespace pa

namespace pa {
typedef __attribute__((ext_vector_type(8))) int intx8_;

__global__ void mfma_probe_fp8mx_kernel(const unsigned char* __restrict__ a,
                                        const unsigned char* __restrict__ b,
                                        float* __restrict__ c,
                                        int sa, int sb) {
  // hypothesis: mfma_scale_f32_16x16x128_f8f6f4 with fmt 0 (e4m3):
  //   A lane l: row = l%16, k bytes 32*(l/16) .. +31   (a: [16][128] rm)
  //   B lane l: col = l%16, same k window              (b: B^T [16][128] rm)
  //   C: col = l%16, row = 4*(l/16)+r  (same as 16x16x32)
  //   scale args: i32 with 4 e8m0 bytes; opsel picks; 0x7F7F7F7F == 1.0
  int lane = threadIdx.x & 63;
  int l16 = lane & 15, lg = lane >> 4;
  intx8_ af = *reinterpret_cast<const intx8_*>(a + l16 * 128 + lg * 32);
  intx8_ bf = *reinterpret_cast<const intx8_*>(b + l16 * 128 + lg * 32);
  floatx4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(af, bf, acc, 0, 0,
                                                         0, sa, 0, sb);
#pragma unroll
  for (int r = 0; r < 4; ++r) c[(lg * 4 + r) * 16 + l16] = acc[r];
}

void mfma_probe_fp8mx(const void* a, const void* bt, float* c, int sa, int sb,
                      hipStream_t s);
void mfma_probe_fp8mx(const void* a, const void* bt, float* c, int sa, int sb,
                      hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe_fp8mx_kernel, dim3(1), dim3(64), 0, s,
                     (const unsigned char*)a, (const unsigned char*)bt, c, sa, sb);
}
}  // namespace pa
