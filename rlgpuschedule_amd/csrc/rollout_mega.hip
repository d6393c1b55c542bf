// Fused whole-rollout "megakernel" — CDNA4 (gfx950).
//
// ONE kernel launch executes the entire T-step rollout (policy forward +
// sampling + DES env transition per step) that the two-kernel engine
// (nn_kernels.hip actor_mlp + env_step.hip) issues as 2*T graph-replayed
// launches. Motivation (docs/ROUND2_MEGAKERNEL.md): at the 81M-steps/s
// headline, every actor->env->actor boundary costs a full wave drain +
// relaunch, and each kernel re-stages obs/state it could have kept local.
//
// Structure:
//   * 256 threads/workgroup = 4 waves; the workgroup owns 16 envs
//     (= the actor tile ROWS) for the WHOLE rollout. grid = ceil(E/16).
//     Envs are independent, so no cross-workgroup sync is ever needed —
//     this is a plain launch, not a cooperative one.
//   * Per step t:
//       phase A (all 4 waves): stage obs[t] tile -> 3-layer f32-MFMA
//         forward (mfma_core_16x64, identical math to actor_mlp_kernel)
//         -> wave-0 masked softmax + inverse-CDF sampling -> write
//         actions/logp/value[t] (global) + actions to LDS.
//       phase B (wave w steps envs [4w, 4w+4) sequentially): the exact
//         env transition from env_device.h, writing obs/mask[t+1]
//         (or obs_last/mask_last at t = T-1) and reward/done[t].
//     __syncthreads() between phases orders LDS AND the global
//     obs[t+1] writes (s_waitcnt vmcnt(0) + barrier) for the next tile
//     staging.
//   * After the loop: value-only forward on obs_last -> last_value
//     (the GAE bootstrap), saving the engine's separate values() call.
//
// Numerics: bit-identical to the two-kernel rollout given the same
// weights and uniforms (same fp32 MFMA code, same env semantics from the
// shared header) — asserted by tests/test_mega_rollout.py.
#include "env_device.h"
#include "nn_common.h"

namespace {

#define MROWS 16     // envs per workgroup (= actor tile rows)
#define MHEADC 32    // head columns (A+1 <= 32)

struct MegaArgs {
  RlsParams p;
  RlsBufs b;                 // b.actions/obs/rewards/done/mask unused here
  int64_t T, H;              // rollout length, hidden (D = p.obs_dim)
  const float *wt1, *b1, *wt2, *b2, *wth, *bh;  // packed fp32 (hip_actor)
  const float* u_all;        // [T,E] uniforms
  float* r_obs;              // [T,E,D]
  uint8_t* r_mask;           // [T,E,A]
  int64_t* r_actions;        // [T,E]
  float* r_logp;             // [T,E]
  float* r_value;            // [T,E]
  float* r_reward;           // [T,E]
  uint8_t* r_done;           // [T,E]
  float* obs_last;           // [E,D]
  uint8_t* mask_last;        // [E,A]
  float* last_value;         // [E]
};

// Actor tile forward for the workgroup's 16 rows starting at row0.
// obs_src: [E,D] global rows for this t. Writes value (always) and, unless
// value_only, samples actions -> global + s_act LDS.
__device__ __forceinline__ void mega_actor_phase(
    const MegaArgs& a, int64_t row0, const float* obs_src,
    const uint8_t* mask_src, const float* u_src, int64_t* act_out,
    float* logp_out, float* val_out, int64_t* s_act,
    float* sobs, float* h1, float* h2, float* shead, int value_only) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int D = (int)a.p.obs_dim, H = (int)a.H, A = (int)a.p.A;
  const int Dp = (D + 15) & ~15;
  const int lda_o = Dp + 4, lda_h = H + 4;
  const int64_t E = a.p.E;

  for (int ix = tid; ix < MROWS * Dp; ix += 256) {
    int r = ix / Dp, d = ix % Dp;
    float vv = 0.0f;
    if (d < D && row0 + r < E) vv = obs_src[(row0 + r) * D + d];
    sobs[r * lda_o + kperm(d, Dp)] = vv;
  }
  __syncthreads();

  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_core_16x64(sobs, lda_o, Dp, a.wt1, H, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        h1[(crow + i) * lda_h + kperm(col, H)] = tanhf(acc[n][i] + a.b1[col]);
      }
  }
  __syncthreads();

  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_core_16x64(h1, lda_h, H, a.wt2, H, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        h2[(crow + i) * lda_h + kperm(col, H)] = tanhf(acc[n][i] + a.b2[col]);
      }
  }
  __syncthreads();

  if (w == 0) {   // head: h2[16,H] @ wth[H,32]
    f32x4 acc[2] = {};
    const int kk = lane >> 4, bc = lane & 15;
    const float* arow = h2 + (lane & 15) * lda_h + kk * (H >> 2);
    const float* bbase = a.wth + (int64_t)4 * (kk * MHEADC + bc);
    for (int k16 = 0; k16 < (H >> 4); ++k16) {
      f32x4 av = *(const f32x4*)(arow + 4 * k16);
      const float* bk = bbase + (int64_t)16 * k16 * MHEADC;
      #pragma unroll
      for (int n = 0; n < 2; ++n) {
        f32x4 bv = *(const f32x4*)(bk + 64 * n);
        #pragma unroll
        for (int s = 0; s < 4; ++s)
          acc[n] = __builtin_amdgcn_mfma_f32_16x16x4f32(av[s], bv[s],
                                                        acc[n], 0, 0, 0);
      }
    }
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 2; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = n * 16 + ccol;
        shead[(crow + i) * (MHEADC + 1) + col] = acc[n][i] + a.bh[col];
      }
  }
  __syncthreads();

  if (w == 0 && lane < MROWS) {
    int64_t row = row0 + lane;
    if (row < E) {
      const float* hrow = shead + lane * (MHEADC + 1);
      val_out[row] = hrow[A];
      if (!value_only) {
        const uint8_t* mrow = mask_src + row * A;
        float mx = -3.0e38f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) mx = fmaxf(mx, hrow[i]);
        float sum = 0.0f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) sum += __expf(hrow[i] - mx);
        float thresh = u_src[row] * sum;
        float s = 0.0f;
        int act = -1;
        for (int i = 0; i < A; ++i) {
          if (!mrow[i]) continue;
          s += __expf(hrow[i] - mx);
          act = i;
          if (s >= thresh) break;
        }
        act_out[row] = act;
        s_act[lane] = act;
        logp_out[row] = (hrow[act] - mx) - __logf(sum);
      }
    }
  }
  __syncthreads();
}

// One env transition (exact env_step_kernel semantics, shared header),
// with per-step output slices.
__device__ void mega_env_step_one(
    const RlsParams& p, const RlsBufs& b, int64_t e, int lane, int64_t a,
    int32_t* s_nf, float* obs_out, uint8_t* mask_out, float* rew_out,
    uint8_t* done_out) {
  EnvView v = make_view(p, b, e);
  v.obs = obs_out;
  v.mask = mask_out;
  Sc s;
  load_sc(b, e, s);
  nf_to_lds(v, s_nf, lane);
  bool did = false;
  float reward = 0.0f;
  if (a >= 0 && a < p.K) {
    if (a < s.pend_count) did = place(v, s, (int)a, lane);
  } else if (p.gandiva && a == p.K + 1) {
    did = do_migrate(v, s);
  } else if (p.gandiva && a == p.K + 2) {
    did = do_timeslice(v, s);
  } else if (p.elastic_resize) {
    int64_t base = p.K + 1 + (p.gandiva ? 2 : 0);
    if (a == base) did = do_shrink(v, s);
    else if (a == base + 1) did = do_expand(v, s);
  }
  if (!did) reward = advance(v, s, lane);
  s.decisions += 1;
  s.reward_sum += reward;
  bool done = (s.cursor == v.job_hi && s.pend_count == 0 && s.run_count == 0)
              || s.decisions >= (int)p.horizon;
  *rew_out = reward;
  *done_out = (uint8_t)done;
  if (done) {
    b.ep_count[e] += 1;
    b.ep_return[e] += s.reward_sum;
    b.ep_jct_sum[e] += s.jct_sum;
    b.ep_completed[e] += s.completed;
    b.ep_decisions[e] += s.decisions;
    reset_env(v, s, lane);
  }
  store_sc(b, e, s);
  write_obs_mask(v, s, lane);
  nf_writeback(v, lane);
}

__global__ __launch_bounds__(256) void mega_rollout_kernel(MegaArgs a) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * MROWS;
  const int64_t E = a.p.E;
  const int D = (int)a.p.obs_dim, H = (int)a.H, A = (int)a.p.A;
  const int Dp = (D + 15) & ~15;
  const int lda_o = Dp + 4, lda_h = H + 4;

  // LDS lifetime aliasing (v2): the actor pipeline only ever needs two
  // tiles alive at once (obs->h1, h1->h2, h2->head), so h2 reuses the
  // obs region and head reuses the h1 region. 57 KB -> ~37 KB per
  // workgroup: occupancy 2 -> 4 workgroups/CU, which doubles the
  // env-phase wave concurrency (the latency-bound part).
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int r1_sz = MROWS * (lda_o > lda_h ? lda_o : lda_h);
  float* sobs = (float*)smem;                       // region 1 (kperm)
  float* h2 = sobs;                                 // region 1 (after L2)
  float* h1 = sobs + r1_sz;                         // region 2 (kperm)
  float* shead = h1;                                // region 2 (after head)
  const int r2_sz = MROWS * (lda_h > (MHEADC + 1) ? lda_h : (MHEADC + 1));
  int64_t* s_act = (int64_t*)(h1 + r2_sz);          // [MROWS]
  int32_t* s_nf = (int32_t*)(s_act + MROWS);        // [4][RLS_MAX_NODES]

  for (int64_t t = 0; t < a.T; ++t) {
    mega_actor_phase(a, row0, a.r_obs + t * E * D, a.r_mask + t * E * A,
                     a.u_all + t * E, a.r_actions + t * E,
                     a.r_logp + t * E, a.r_value + t * E, s_act,
                     sobs, h1, h2, shead, /*value_only=*/0);
    // phase B: wave w owns envs [row0+4w, row0+4w+4)
    const bool last = (t == a.T - 1);
    float* obs_nxt = last ? a.obs_last : a.r_obs + (t + 1) * E * D;
    uint8_t* mask_nxt = last ? a.mask_last : a.r_mask + (t + 1) * E * A;
    for (int i = 0; i < 4; ++i) {
      int64_t e = row0 + w * 4 + i;
      if (e < E) {
        mega_env_step_one(a.p, a.b, e, lane, (int64_t)s_act[w * 4 + i],
                          s_nf + w * RLS_MAX_NODES,
                          obs_nxt + e * D, mask_nxt + e * A,
                          a.r_reward + t * E + e, a.r_done + t * E + e);
      }
    }
    __syncthreads();   // orders obs[t+1] global writes before next staging
  }
  // GAE bootstrap: value-only forward on the final observation
  mega_actor_phase(a, row0, a.obs_last, a.mask_last, a.u_all,
                   a.r_actions, a.r_logp, a.last_value, s_act,
                   sobs, h1, h2, shead, /*value_only=*/1);
}

}  // namespace

extern "C" int rlsched_mega_rollout(const int64_t* ip, const uint64_t* bufs,
                                    void* stream) {
  MegaArgs a;
  unpack(ip, bufs, a.p, a.b);     // first 17 params / 34 bufs = env ABI
  a.T = ip[17];
  a.H = ip[18];
  if (a.p.A + 1 > MHEADC || (a.H & 63) || (a.p.N > RLS_MAX_NODES))
    return -2;
  const uint64_t* x = bufs + RLS_NUM_BUFS;
  a.wt1 = (const float*)x[0]; a.b1 = (const float*)x[1];
  a.wt2 = (const float*)x[2]; a.b2 = (const float*)x[3];
  a.wth = (const float*)x[4]; a.bh = (const float*)x[5];
  a.u_all = (const float*)x[6];
  a.r_obs = (float*)x[7];
  a.r_mask = (uint8_t*)x[8];
  a.r_actions = (int64_t*)x[9];
  a.r_logp = (float*)x[10];
  a.r_value = (float*)x[11];
  a.r_reward = (float*)x[12];
  a.r_done = (uint8_t*)x[13];
  a.obs_last = (float*)x[14];
  a.mask_last = (uint8_t*)x[15];
  a.last_value = (float*)x[16];
  const int D = (int)a.p.obs_dim, H = (int)a.H;
  const int Dp = (D + 15) & ~15;
  const int lda_o = Dp + 4, lda_h = H + 4;
  const int r1 = MROWS * (lda_o > lda_h ? lda_o : lda_h);
  const int r2 = MROWS * (lda_h > (MHEADC + 1) ? lda_h : (MHEADC + 1));
  size_t lds = sizeof(float) * (r1 + r2)
             + sizeof(int64_t) * MROWS
             + sizeof(int32_t) * 4 * RLS_MAX_NODES;
  if (lds > 64 * 1024) {   // H=512 tile: opt in to the 160 KB LDS budget
    static bool lds_opt_in = false;
    if (!lds_opt_in) {
      hipError_t rc = hipFuncSetAttribute(
          (const void*)mega_rollout_kernel,
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
      if (rc != hipSuccess) return (int)rc;
      lds_opt_in = true;
    }
  }
  dim3 grid((unsigned)((a.p.E + MROWS - 1) / MROWS));
  hipLaunchKernelGGL(mega_rollout_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, a);
  return (int)hipGetLastError();
}

// Build provenance (see ops/build.py): hash of this lib's sources.
#ifndef RLS_SOURCE_HASH
#define RLS_SOURCE_HASH "unset"
#endif
extern "C" const char* rlsched_source_hash() {
  return "RLSHASH:" RLS_SOURCE_HASH;
}
