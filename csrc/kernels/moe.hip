// MoE routing kernels for gfx950: fused top-k gate + deterministic
// capacity slot assignment.
//
// Reference role: paddle/phi/kernels/gpu/{assign_pos,number_count,
// limit_by_capacity,prune_gate_by_capacity}_kernel.cu + the gate math in
// incubate/distributed/models/moe/gate/gshard_gate.py -- re-derived as
// two wave64 kernels instead of the reference's kernel-per-step chain:
//   * gate: one wave per token (E <= 64: one expert logit per lane) does
//     softmax + top-k + load-balance stats in registers
//   * assign: one wave per (expert, k) walks tokens in 64-chunks using
//     __ballot + popc prefix -- positions follow token order exactly like
//     the reference's cumsum (deterministic capacity dropping), no sort
#include "common.h"
#include "api.h"

namespace pa {

// logits [T, E] fp32 (E <= 64), k <= 4
// outputs: topv [T,k] f32 (softmax probs), topi [T,k] i32,
//          me [E] f32 (+= mean prob), ce [E] f32 (+= top1 counts / T)
__global__ void moe_gate_kernel(const float* __restrict__ logits,
                                float* __restrict__ topv, int* __restrict__ topi,
                                float* __restrict__ me, float* __restrict__ ce,
                                long long T, int E, int K) {
  const long long t = ((long long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  if (t >= T) return;
  float lg = (lane < E) ? logits[t * E + lane] : -1e30f;
  // softmax over the wave
  float mx = wave_reduce(lg, MaxOp());
  float ex = (lane < E) ? __expf(lg - mx) : 0.f;
  float den = wave_reduce(ex, SumOp());
  float p = ex / den;
  if (lane < E) atomicAdd(&me[lane], p / (float)T);
  float pw = p;
  for (int kk = 0; kk < K; ++kk) {
    float best = wave_reduce(pw, MaxOp());
    // first lane holding the max wins
    unsigned long long win = __ballot(pw == best);
    int win_lane = (int)__ffsll((long long)win) - 1;
    if (lane == 0) {
      topv[t * K + kk] = best;
      topi[t * K + kk] = win_lane;
      if (kk == 0) atomicAdd(&ce[win_lane], 1.f / (float)T);
    }
    if (lane == win_lane) pw = -1.f;   // remove from next round
  }
}

// topi [T, k] i32 -> slot_of [T, k] i32 (expert*cap + pos, or -1 dropped),
// counts [E] i32.  grid = E * K waves; wave w handles (expert e, slot kk).
__global__ void moe_assign_kernel(const int* __restrict__ topi,
                                  int* __restrict__ slot_of,
                                  int* __restrict__ counts,
                                  long long T, int E, int K, int cap) {
  const int w = ((int)blockIdx.x * (int)blockDim.x + (int)threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  if (w >= E * K) return;
  const int e = w / K, kk = w % K;
  int base = 0;
  for (long long t0 = 0; t0 < T; t0 += 64) {
    long long t = t0 + lane;
    bool match = (t < T) && (topi[t * K + kk] == e);
    unsigned long long mask = __ballot(match);
    if (match) {
      int pos = base + __popcll(mask & ((1ull << lane) - 1));
      slot_of[t * K + kk] = (pos < cap) ? (e * cap + pos) : -1;
    }
    base += __popcll(mask);
  }
  if (lane == 0) atomicAdd(&counts[e], base);
}

void moe_gate_topk(const float* logits, float* topv, int* topi, float* me,
                   float* ce, int64_t t, int64_t e, int64_t k, hipStream_t s) {
  long long waves = t;
  dim3 g((unsigned)hmin<long long>((waves * 64 + 255) / 256, 1 << 30));
  hipLaunchKernelGGL(moe_gate_kernel, g, dim3(256), 0, s, logits, topv, topi,
                     me, ce, t, (int)e, (int)k);
}

void moe_assign_slots(const int* topi, int* slot_of, int* counts, int64_t t,
                      int64_t e, int64_t k, int64_t cap, hipStream_t s) {
  long long waves = e * k;
  dim3 g((unsigned)((waves * 64 + 255) / 256));
  hipLaunchKernelGGL(moe_assign_kernel, g, dim3(256), 0, s, topi, slot_of,
                     counts, t, (int)e, (int)k, (int)cap);
}

}  // namespace pa
