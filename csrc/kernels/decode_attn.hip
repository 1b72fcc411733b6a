// Paged-KV single-token decode attention for gfx950 (serving path).
//
// Reference role parity: paddle/phi/kernels/fusion/gpu/
// masked_multihead_attention_kernel.cu + block_multi_head_attention
// (paged "block" KV cache) -- re-derived for wave64:
//   * one 256-thread block per (batch, q-head); GQA via head mapping
//   * KV streamed from a paged cache: k/v_cache [nblocks, bs, hkv, D],
//     block_table [B, max_blocks] -- HBM-bound, 16B/lane loads
//   * 16 lanes per position (D<=128: 8 elems/lane), 4 pos/wave/step,
//     online softmax in registers, cross-wave merge via LDS
#include "common.h"
#include "api.h"

namespace pa {

template <int D>
__launch_bounds__(256)
__global__ void decode_attn_kernel(const short* __restrict__ qg,
                                   const short* __restrict__ kcache,
                                   const short* __restrict__ vcache,
                                   const int* __restrict__ block_table,
                                   const int* __restrict__ seq_lens,
                                   short* __restrict__ og,
                                   int B, int H, int HKV, int bs,
                                   int max_blocks, float scale,
                                   long long blk_str, long long pos_str,
                                   long long head_str) {
  const int b = blockIdx.x / H;
  const int h = blockIdx.x % H;
  const int hkv = h / (H / HKV);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l16 = lane & 15;   // element group within D
  const int lp = lane >> 4;    // 0..3: position slot within wave
  const int S = seq_lens[b];

  // q for this (b,h): D elems; 16-lane group covers D -> E = D/16 per lane
  constexpr int E = D / 16;
  float qv[E];
  {
    const short* qp = qg + ((long long)b * H + h) * D;
#pragma unroll
    for (int i = 0; i < E; ++i) qv[i] = bf2f(qp[l16 * E + i]);
  }

  float m = -1e30f, l = 0.f;  // finite sentinel: -ffast-math breaks +-inf
  float oacc[E];
#pragma unroll
  for (int i = 0; i < E; ++i) oacc[i] = 0.f;

  // each wave processes 4 positions per step; 4 waves -> 16 pos/block/step.
  // Software-pipelined: K/V for step t+1 load while step t computes --
  // the serial position walk was latency-bound (back-to-back dependent
  // 16 B loads per step).
  auto load_kv = [&](int pos0, bool& ok, shortx8& kv8, shortx8& vv8) {
    int pos = pos0 + lp;
    ok = pos < S;
    int phys = 0, off = 0;
    if (ok) {
      phys = block_table[(long long)b * max_blocks + pos / bs];
      off = pos % bs;
    }
    const long long base = (long long)phys * blk_str +
                           (long long)off * pos_str +
                           (long long)hkv * head_str + l16 * E;
    if (ok) {
      if (E == 8) {
        kv8 = *reinterpret_cast<const shortx8*>(kcache + base);
        vv8 = *reinterpret_cast<const shortx8*>(vcache + base);
      } else {
        shortx4 k4 = *reinterpret_cast<const shortx4*>(kcache + base);
        shortx4 v4 = *reinterpret_cast<const shortx4*>(vcache + base);
#pragma unroll
        for (int i = 0; i < 4; ++i) { kv8[i] = k4[i]; vv8[i] = v4[i]; }
      }
    }
  };

  bool okc = false, okn = false;
  shortx8 kc8, vc8, kn8, vn8;
  const int p_start = wid * 4;
  load_kv(p_start, okc, kc8, vc8);
  for (int pos0 = p_start; pos0 < S; pos0 += 16) {
    if (pos0 + 16 < S) load_kv(pos0 + 16, okn, kn8, vn8);
    else okn = false;
    bool ok = okc;
    float sc = 0.f;
    if (ok) {
#pragma unroll
      for (int i = 0; i < E; ++i) sc += qv[i] * bf2f(kc8[i]);
    }
    // reduce over the 16-lane group
    sc = group16_reduce(sc, SumOp());
    sc = ok ? sc * scale : -1e30f;
    // online softmax across the 4 position-slots handled by this wave-step
    float m_new = fmaxf(m, sc);
    float corr = __expf(m - m_new);       // underflows to 0 for the sentinel
    float p = ok ? __expf(sc - m_new) : 0.f;
    l = l * corr + p;
    m = m_new;
#pragma unroll
    for (int i = 0; i < E; ++i) {
      float vvf = ok ? bf2f(vc8[i]) : 0.f;
      oacc[i] = oacc[i] * corr + p * vvf;
    }
    okc = okn; kc8 = kn8; vc8 = vn8;
  }

  // ---- merge the 16 position-slot partials (4 waves x 4 slots) ----------
  // each (wid, lp) slot has (m, l, oacc[8]) per l16 group
  __shared__ float sm[16], sl[16];
  __shared__ float so[16][D];
  const int slot_id = wid * 4 + lp;
  if (l16 == 0) {
    sm[slot_id] = m;
    sl[slot_id] = l;
  }
#pragma unroll
  for (int i = 0; i < E; ++i) so[slot_id][l16 * E + i] = oacc[i];
  __syncthreads();
  if (tid < 64) {
    // wave 0 merges 16 slots
    float gm = -1e30f;
#pragma unroll
    for (int s2 = 0; s2 < 16; ++s2) gm = fmaxf(gm, sm[s2]);
    float gl = 0.f;
    float out[E];
#pragma unroll
    for (int i = 0; i < E; ++i) out[i] = 0.f;
#pragma unroll
    for (int s2 = 0; s2 < 16; ++s2) {
      float w = (sl[s2] > 0.f) ? __expf(sm[s2] - gm) : 0.f;
      gl += sl[s2] * w;
#pragma unroll
      for (int i = 0; i < E; ++i) out[i] += w * so[s2][l16 * E + i];
    }
    if (lane < 16) {
      float inv = gl > 0.f ? 1.f / gl : 0.f;
      short* op = og + ((long long)b * H + h) * D;
#pragma unroll
      for (int i = 0; i < E; ++i) op[l16 * E + i] = f2bf(out[i] * inv);
    }
  }
}

void decode_attention(const void* q, const void* kcache, const void* vcache,
                      const int* block_table, const int* seq_lens, void* o,
                      int64_t b, int64_t h, int64_t hkv, int64_t bs,
                      int64_t max_blocks, int64_t dh, float scale,
                      int64_t blk_str, int64_t pos_str, int64_t head_str,
                      hipStream_t s) {
  dim3 grid((unsigned)(b * h));
  if (blk_str == 0) {   // default paged layout [nblocks, bs, HKV, D]
    blk_str = bs * hkv * dh;
    pos_str = hkv * dh;
    head_str = dh;
  }
  if (dh == 128)
    hipLaunchKernelGGL((decode_attn_kernel<128>), grid, dim3(256), 0, s,
                       (const short*)q, (const short*)kcache, (const short*)vcache,
                       block_table, seq_lens, (short*)o, (int)b, (int)h,
                       (int)hkv, (int)bs, (int)max_blocks, scale,
                       blk_str, pos_str, head_str);
  else
    hipLaunchKernelGGL((decode_attn_kernel<64>), grid, dim3(256), 0, s,
                       (const short*)q, (const short*)kcache, (const short*)vcache,
                       block_table, seq_lens, (short*)o, (int)b, (int)h,
                       (int)hkv, (int)bs, (int)max_blocks, scale,
                       blk_str, pos_str, head_str);
}

}  // namespace pa
