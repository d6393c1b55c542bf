// Hand-written CDNA4 kernels for the RL hot path: GAE scan + fused MLP actor.
//
// actor_mlp_kernel — ONE launch per rollout step computes, for all E envs:
//     h1 = tanh(obs @ W1t + b1)      [E,H]   (MFMA f32 16x16x4)
//     h2 = tanh(h1 @ W2t + b2)       [E,H]
//     head = h2 @ Wht + bh           [E,32]  (cols [0,A) logits, col A value)
//     masked softmax -> inverse-CDF sample with caller-provided uniforms
//   Layout: 16 env rows per 256-thread workgroup (4 waves); each wave owns
//   64 output columns per chunk; activations live in LDS (padded stride
//   H+1 to break bank conflicts); weights stream from global and stay
//   L2-resident (W2 fp32 = 256 KB/XCD-L2). MFMA f32_16x16x4 is exact f32
//   at the 157 TF vector rate (cdna_hip_programming.md §3) — numerics match
//   the torch fp32 reference to reassociation tolerance.
//   Weight tensors are the TRANSPOSED copies maintained by
//   models/hip_actor.py from the torch parameters ([in,out] layout so the
//   B-fragment read B[k][n] is coalesced: 16 consecutive lanes read 16
//   consecutive columns).
//
// gae_kernel — segmented reverse scan over [T,E]: one thread per env,
//   time-step loads coalesced across the wave (layout [t][e]).
//
// Numerics tests: tests/test_nn_kernels.py vs torch fp32 reference.

#include "nn_common.h"

namespace {

// ---------------------------------------------------------------- GAE scan
struct GaeArgs {
  int64_t T, E;
  float gamma, lam;
  const float* reward;      // [T*E]
  const float* value;       // [T*E]
  const uint8_t* done;      // [T*E]
  const float* last_value;  // [E]
  float* adv;               // [T*E]
  float* ret;               // [T*E]
};

__global__ __launch_bounds__(256) void gae_kernel(GaeArgs a) {
  int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (e >= a.E) return;
  float lastg = 0.0f;
  float next_v = a.last_value[e];
  for (int64_t t = a.T - 1; t >= 0; --t) {
    int64_t ix = t * a.E + e;
    float nd = 1.0f - (float)a.done[ix];
    float v = a.value[ix];
    float delta = a.reward[ix] + a.gamma * next_v * nd - v;
    lastg = delta + a.gamma * a.lam * nd * lastg;
    a.adv[ix] = lastg;
    a.ret[ix] = lastg + v;
    next_v = v;
  }
}

// ------------------------------------------------------- fused MLP actor
struct ActorArgs {
  int64_t E, D, H, A;       // envs, obs_dim, hidden, num_actions
  int64_t value_only;       // 1: only write value (bootstrap pass)
  const float* obs;         // [E,D]
  const uint8_t* mask;      // [E,A]
  const float* u;           // [E] uniforms (ignored if value_only)
  const float* wt1;         // [D,H] transposed W1
  const float* b1;          // [H]
  const float* wt2;         // [H,H]
  const float* b2;          // [H]
  const float* wth;         // [H,32] logits cols [0,A), value col A, rest 0
  const float* bh;          // [32]
  int64_t* actions;         // [E]
  float* logp;              // [E]
  float* value;             // [E]
};

#define ROWS 16      // env rows per workgroup
#define HEADC 32     // head columns (A+1 <= 32)

// K-permutation for float4 fragment reads: element k of a K-vector is
// stored at kperm(k) = (k%4)*(K/4) + k/4 (bijective, K % 16 == 0). A lane
// with MFMA k-subindex kk then reads its four k's for a 16-deep macro-step
// as ONE aligned float4 at [kk*(K/4) + 4*k16]. Weights are packed to match
// by models/hip_actor.py: wt[K,H] -> view(K/16,4(s),4(kk),H)
// .permute(0,2,3,1) so a lane's B values for the macro-step are one float4
// at (((k16*4+kk)*H + col)*4) — 16 consecutive lanes read 16 consecutive
// float4s (fully coalesced 1 KiB per instruction).
// One 64-column GEMM chunk for one wave: C[16 x 64] += A[16 x K] * B[K x 64].
// A: LDS tile in kperm layout, row stride lda. B: packed global (see above),
// logical row length ldb. Output written through bias(+tanh) epilogue into
// LDS in kperm layout of the NEXT layer (perm_out) or plain (head).
__device__ __forceinline__ void gemm_chunk_16x64(
    const float* __restrict__ a_lds, int lda, int K,
    const float* __restrict__ b_pk, int ldb, int col0,
    const float* __restrict__ bias, bool do_tanh,
    float* out_lds, int ldo, int lane, int perm_out, int Kout) {
  f32x4 acc[4] = {};
  const int ar = lane & 15;        // A row
  const int kk = lane >> 4;        // k sub-index (0..3)
  const int bc = lane & 15;        // B col within 16-group
  const float* arow = a_lds + ar * lda + kk * (K >> 2);
  const float* bbase = b_pk + (int64_t)4 * (kk * ldb + col0 + bc);
  const int nk16 = K >> 4;
  for (int k16 = 0; k16 < nk16; ++k16) {
    f32x4 av = *(const f32x4*)(arow + 4 * k16);
    const float* bk = bbase + (int64_t)16 * k16 * ldb;
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      f32x4 bv = *(const f32x4*)(bk + 64 * n);
      #pragma unroll
      for (int s = 0; s < 4; ++s)
        acc[n] = __builtin_amdgcn_mfma_f32_16x16x4f32(av[s], bv[s], acc[n],
                                                      0, 0, 0);
    }
  }
  // C/D map (16x16): col = lane&15, row = (lane>>4)*4 + i
  const int crow = (lane >> 4) * 4;
  const int ccol = lane & 15;
  #pragma unroll
  for (int n = 0; n < 4; ++n) {
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int col = col0 + n * 16 + ccol;
      float vv = acc[n][i] + bias[col];
      if (do_tanh) vv = tanhf(vv);
      int oc = perm_out ? kperm(col, Kout) : col;
      out_lds[(crow + i) * ldo + oc] = vv;
    }
  }
}

__global__ __launch_bounds__(256) void actor_mlp_kernel(ActorArgs p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * ROWS;
  const int D = (int)p.D, H = (int)p.H, A = (int)p.A;
  const int Dp = (D + 15) & ~15;            // K-padded obs dim
  // row strides: multiples of 4 floats (16B alignment for f32x4 reads) and
  // != 0 mod 64 dwords so 16 rows hit 16 distinct bank groups
  const int lda_o = Dp + 4, lda_h = H + 4;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* sobs = (float*)smem;                       // [ROWS][Dp+4] kperm
  float* h1 = sobs + ROWS * lda_o;                  // [ROWS][H+4] kperm
  float* h2 = h1 + ROWS * lda_h;                    // [ROWS][H+4] kperm
  float* shead = h2 + ROWS * lda_h;                 // [ROWS][HEADC+1]

  // stage obs tile into kperm layout (zero-pad rows past E and d >= D)
  for (int ix = tid; ix < ROWS * Dp; ix += 256) {
    int r = ix / Dp, d = ix % Dp;
    float vv = 0.0f;
    if (d < D && row0 + r < p.E) vv = p.obs[(row0 + r) * D + d];
    sobs[r * lda_o + kperm(d, Dp)] = vv;
  }
  __syncthreads();

  // layer 1: obs[16,Dp] @ wt1p[Dp,H]
  for (int c0 = w * 64; c0 < H; c0 += 256)
    gemm_chunk_16x64(sobs, lda_o, Dp, p.wt1, H, c0, p.b1, true,
                     h1, lda_h, lane, 1, H);
  __syncthreads();

  // layer 2: h1[16,H] @ wt2p[H,H]
  for (int c0 = w * 64; c0 < H; c0 += 256)
    gemm_chunk_16x64(h1, lda_h, H, p.wt2, H, c0, p.b2, true,
                     h2, lda_h, lane, 1, H);
  __syncthreads();

  // head: h2[16,H] @ wthp[H,32] — wave 0 only (tiny)
  if (w == 0) {
    f32x4 acc[2] = {};
    const int kk = lane >> 4, bc = lane & 15;
    const float* arow = h2 + (lane & 15) * lda_h + kk * (H >> 2);
    const float* bbase = p.wth + (int64_t)4 * (kk * HEADC + bc);
    for (int k16 = 0; k16 < (H >> 4); ++k16) {
      f32x4 av = *(const f32x4*)(arow + 4 * k16);
      const float* bk = bbase + (int64_t)16 * k16 * HEADC;
      #pragma unroll
      for (int n = 0; n < 2; ++n) {
        f32x4 bv = *(const f32x4*)(bk + 64 * n);
        #pragma unroll
        for (int s = 0; s < 4; ++s)
          acc[n] = __builtin_amdgcn_mfma_f32_16x16x4f32(av[s], bv[s], acc[n],
                                                        0, 0, 0);
      }
    }
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 2; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = n * 16 + ccol;
        shead[(crow + i) * (HEADC + 1) + col] = acc[n][i] + p.bh[col];
      }
  }
  __syncthreads();

  // per-row masked softmax + inverse-CDF sample: lane r of wave 0 owns row r
  if (w == 0 && lane < ROWS) {
    int64_t row = row0 + lane;
    if (row < p.E) {
      const float* hrow = shead + lane * (HEADC + 1);
      p.value[row] = hrow[A];
      if (!p.value_only) {
        const uint8_t* mrow = p.mask + row * A;
        float mx = -3.0e38f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) mx = fmaxf(mx, hrow[i]);
        float sum = 0.0f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) sum += __expf(hrow[i] - mx);
        float thresh = p.u[row] * sum;
        float s = 0.0f;
        int act = -1;
        for (int i = 0; i < A; ++i) {
          if (!mrow[i]) continue;
          s += __expf(hrow[i] - mx);
          act = i;                 // last valid as fallback
          if (s >= thresh) break;
        }
        p.actions[row] = act;
        p.logp[row] = (hrow[act] - mx) - __logf(sum);
      }
    }
  }
}

}  // namespace

extern "C" int rlsched_gae(const int64_t* ip, const double* fp,
                           const uint64_t* bufs, void* stream) {
  GaeArgs a;
  a.T = ip[0]; a.E = ip[1];
  a.gamma = (float)fp[0]; a.lam = (float)fp[1];
  a.reward = (const float*)bufs[0];
  a.value = (const float*)bufs[1];
  a.done = (const uint8_t*)bufs[2];
  a.last_value = (const float*)bufs[3];
  a.adv = (float*)bufs[4];
  a.ret = (float*)bufs[5];
  dim3 grid((unsigned)((a.E + 255) / 256));
  hipLaunchKernelGGL(gae_kernel, grid, dim3(256), 0, (hipStream_t)stream, a);
  return (int)hipGetLastError();
}

extern "C" int rlsched_actor_mlp(const int64_t* ip, const uint64_t* bufs,
                                 void* stream) {
  ActorArgs p;
  p.E = ip[0]; p.D = ip[1]; p.H = ip[2]; p.A = ip[3]; p.value_only = ip[4];
  if (p.H % 64 || p.A + 1 > HEADC) return -1;
  p.obs = (const float*)bufs[0];
  p.mask = (const uint8_t*)bufs[1];
  p.u = (const float*)bufs[2];
  p.wt1 = (const float*)bufs[3];
  p.b1 = (const float*)bufs[4];
  p.wt2 = (const float*)bufs[5];
  p.b2 = (const float*)bufs[6];
  p.wth = (const float*)bufs[7];
  p.bh = (const float*)bufs[8];
  p.actions = (int64_t*)bufs[9];
  p.logp = (float*)bufs[10];
  p.value = (float*)bufs[11];
  const int Dp = ((int)p.D + 15) & ~15;
  size_t lds = sizeof(float) * (ROWS * (Dp + 4) + 2 * ROWS * ((int)p.H + 4)
                                + ROWS * (HEADC + 1));
  dim3 grid((unsigned)((p.E + ROWS - 1) / ROWS));
  hipLaunchKernelGGL(actor_mlp_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

// ----------------------------------------------------------- PPO loss head
// Fused forward+backward of the PPO objective given policy outputs:
//   softmax over masked logits (mask baked into logits as -1e9),
//   ratio-clip policy loss, clipped value loss, entropy bonus.
// One pass per sample computes loss partial-sums AND dlogits/dvalue.
// Replaces ~20 torch elementwise/softmax/gather kernels per minibatch.
namespace {

struct PpoLossArgs {
  int64_t B, A;
  float clip, vf_coef, ent_coef, inv_n;  // inv_n = 1/B (mean scaling)
  const float* logits;     // [B,A] (masked)
  const float* value;      // [B]
  const int64_t* actions;  // [B]
  const float* old_logp;   // [B]
  const float* old_value;  // [B]
  const float* adv;        // [B]
  const float* ret;        // [B]
  float* dlogits;          // [B,A]
  float* dvalue;           // [B]
  float* stats;            // [4]: pi_loss, v_loss, entropy, clipfrac (sums)
};

__global__ __launch_bounds__(256) void ppo_loss_kernel(PpoLossArgs a) {
  const int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int A = (int)a.A;
  float s_pi = 0.f, s_v = 0.f, s_ent = 0.f, s_cf = 0.f;
  for (int64_t i = i0; i < a.B; i += (int64_t)gridDim.x * blockDim.x) {
    const float* lg = a.logits + i * A;
    float* dlg = a.dlogits + i * A;
    const int act = (int)a.actions[i];
    // softmax (numerically stable)
    float mx = -3.0e38f;
    for (int j = 0; j < A; ++j) mx = fmaxf(mx, lg[j]);
    float sum = 0.f;
    for (int j = 0; j < A; ++j) sum += __expf(lg[j] - mx);
    const float lsum = __logf(sum);
    const float logp_a = (lg[act] - mx) - lsum;
    // entropy H = -sum p*logp
    float H = 0.f;
    for (int j = 0; j < A; ++j) {
      float lpj = (lg[j] - mx) - lsum;
      H -= __expf(lpj) * lpj;
    }
    // ratio clip (torch.min ties -> first arg, matches agents/ppo.py)
    const float advi = a.adv[i];
    const float ratio = __expf(logp_a - a.old_logp[i]);
    const float rcl = fminf(fmaxf(ratio, 1.f - a.clip), 1.f + a.clip);
    const float s1 = ratio * advi, s2 = rcl * advi;
    const float active = (s1 <= s2) ? 1.f : 0.f;
    s_pi += -fminf(s1, s2);
    s_ent += H;
    s_cf += (fabsf(ratio - 1.f) > a.clip) ? 1.f : 0.f;
    // d(pi_loss)/d(logp_a) = -adv*ratio when unclipped branch is active
    const float g_logp = -advi * ratio * active * a.inv_n;
    // dlogits: g_logp*(delta - p) + ent_coef * p * (logp + H) / n
    for (int j = 0; j < A; ++j) {
      float lpj = (lg[j] - mx) - lsum;
      float pj = __expf(lpj);
      float g = g_logp * (((j == act) ? 1.f : 0.f) - pj)
              + a.ent_coef * pj * (lpj + H) * a.inv_n;
      dlg[j] = g;
    }
    // clipped value loss (torch.max ties -> first arg)
    const float v = a.value[i], ov = a.old_value[i], rt = a.ret[i];
    const float vcl = ov + fminf(fmaxf(v - ov, -a.clip), a.clip);
    const float e1 = v - rt, e2 = vcl - rt;
    const float q1 = e1 * e1, q2 = e2 * e2;
    s_v += 0.5f * fmaxf(q1, q2);
    float dv;
    if (q1 >= q2) dv = e1;
    else dv = (fabsf(v - ov) < a.clip) ? e2 : 0.f;
    a.dvalue[i] = a.vf_coef * dv * a.inv_n;
  }
  // block-reduce the stats then one atomic per block
  __shared__ float red[4][256 / WAVE];
  float vals[4] = {s_pi, s_v, s_ent, s_cf};
  const int lane = threadIdx.x & (WAVE - 1), w = threadIdx.x >> 6;
  #pragma unroll
  for (int k = 0; k < 4; ++k) {
    float x = vals[k];
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
    if (lane == 0) red[k][w] = x;
  }
  __syncthreads();
  if (threadIdx.x < 4) {
    float x = 0.f;
    for (int q = 0; q < 256 / WAVE; ++q) x += red[threadIdx.x][q];
    atomicAdd(&a.stats[threadIdx.x], x);
  }
}

}  // namespace

extern "C" int rlsched_ppo_loss(const int64_t* ip, const double* fp,
                                const uint64_t* bufs, void* stream) {
  PpoLossArgs a;
  a.B = ip[0]; a.A = ip[1];
  a.clip = (float)fp[0]; a.vf_coef = (float)fp[1];
  a.ent_coef = (float)fp[2]; a.inv_n = (float)fp[3];
  a.logits = (const float*)bufs[0];
  a.value = (const float*)bufs[1];
  a.actions = (const int64_t*)bufs[2];
  a.old_logp = (const float*)bufs[3];
  a.old_value = (const float*)bufs[4];
  a.adv = (const float*)bufs[5];
  a.ret = (const float*)bufs[6];
  a.dlogits = (float*)bufs[7];
  a.dvalue = (float*)bufs[8];
  a.stats = (float*)bufs[9];
  int grid = (int)((a.B + 255) / 256);
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(ppo_loss_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, a);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------- V-trace scan
// IMPALA off-policy correction (Espeholt et al. 2018), same segmented
// reverse-scan structure as gae_kernel: one thread per env, coalesced
// [t][e] loads. Produces vs targets and policy-gradient advantages.
namespace {

struct VtraceArgs {
  int64_t T, E;
  float gamma, rho_bar, c_bar, lam;
  const float* reward;      // [T*E]
  const float* value;       // [T*E] V(x_t) under CURRENT params
  const uint8_t* done;      // [T*E]
  const float* last_value;  // [E]
  const float* lp_cur;      // [T*E] log pi(a_t) current
  const float* lp_beh;      // [T*E] log mu(a_t) behavior
  float* vs;                // [T*E] out
  float* pg_adv;            // [T*E] out
};

__global__ __launch_bounds__(256) void vtrace_kernel(VtraceArgs a) {
  int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (e >= a.E) return;
  float acc = 0.0f;                  // vs_{t+1} - V_{t+1}
  float next_v = a.last_value[e];    // V_{t+1}
  float vs_next = a.last_value[e];   // vs_{t+1}
  for (int64_t t = a.T - 1; t >= 0; --t) {
    int64_t ix = t * a.E + e;
    float nd = 1.0f - (float)a.done[ix];
    float v = a.value[ix];
    float r = a.reward[ix];
    float ratio = __expf(a.lp_cur[ix] - a.lp_beh[ix]);
    float rho = fminf(a.rho_bar, ratio);
    float c = a.lam * fminf(a.c_bar, ratio);
    float delta = rho * (r + a.gamma * next_v * nd - v);
    acc = delta + a.gamma * c * nd * acc;
    float vst = v + acc;
    a.vs[ix] = vst;
    a.pg_adv[ix] = rho * (r + a.gamma * vs_next * nd - v);
    vs_next = vst;
    next_v = v;
  }
}

}  // namespace

extern "C" int rlsched_vtrace(const int64_t* ip, const double* fp,
                              const uint64_t* bufs, void* stream) {
  VtraceArgs a;
  a.T = ip[0]; a.E = ip[1];
  a.gamma = (float)fp[0]; a.rho_bar = (float)fp[1];
  a.c_bar = (float)fp[2]; a.lam = (float)fp[3];
  a.reward = (const float*)bufs[0];
  a.value = (const float*)bufs[1];
  a.done = (const uint8_t*)bufs[2];
  a.last_value = (const float*)bufs[3];
  a.lp_cur = (const float*)bufs[4];
  a.lp_beh = (const float*)bufs[5];
  a.vs = (float*)bufs[6];
  a.pg_adv = (float*)bufs[7];
  dim3 grid((unsigned)((a.E + 255) / 256));
  hipLaunchKernelGGL(vtrace_kernel, grid, dim3(256), 0, (hipStream_t)stream, a);
  return (int)hipGetLastError();
}

// ---------------------------------------------------------- fused GRU actor
// One launch per rollout step for the recurrent policy (GRUPolicy):
//   x  = tanh(obs @ Win + bin)
//   r  = sigmoid(x@Wir + bir + h@Whr + bhr)        (torch GRUCell order
//   z  = sigmoid(x@Wiz + biz + h@Whz + bhz)         r, z, n)
//   n  = tanh(x@Win2 + bin2 + r * (h@Whn + bhn))
//   h' = (1-z)*n + z*h ;  head = h' @ Wht + bh ; masked sample
// Gate GEMMs run chunk-wise (64 cols x 6 gate accumulators in registers),
// so the 3H pre-activations are never materialized. h is zeroed for envs
// whose previous transition ended an episode (done_prev).
namespace {

struct GruActorArgs {
  int64_t E, D, H, A;
  int64_t value_only;
  const float* obs;        // [E,D]
  const uint8_t* mask;     // [E,A]
  const float* u;          // [E]
  const uint8_t* done_prev;  // [E] or null: zero h where set
  const float* hx;         // [E,H] in
  float* hx_out;           // [E,H] out (may alias hx)
  const float* wtin;       // packed [Dp,H]
  const float* bin;        // [H]
  const float* wih;        // packed [H,3H] (weight_ih.t())
  const float* bih;        // [3H]
  const float* whh;        // packed [H,3H]
  const float* bhh;        // [3H]
  const float* wth;        // packed [H,HEADC]
  const float* bh;         // [HEADC]
  int64_t* actions;
  float* logp;
  float* value;
};

__device__ __forceinline__ float sigmoidf_(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

__global__ __launch_bounds__(256) void gru_actor_kernel(GruActorArgs p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * ROWS;
  const int D = (int)p.D, H = (int)p.H, A = (int)p.A;
  const int Dp = (D + 15) & ~15;
  const int lda_o = Dp + 4, lda_h = H + 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* sobs = (float*)smem;                 // [ROWS][Dp+4] kperm
  float* sx = sobs + ROWS * lda_o;            // [ROWS][H+4] kperm
  float* shx = sx + ROWS * lda_h;             // [ROWS][H+4] kperm
  float* shn = shx + ROWS * lda_h;            // [ROWS][H+4] kperm (h')
  float* shead = shn + ROWS * lda_h;          // [ROWS][HEADC+1]

  for (int ix = tid; ix < ROWS * Dp; ix += 256) {
    int r = ix / Dp, d = ix % Dp;
    float vv = 0.0f;
    if (d < D && row0 + r < p.E) vv = p.obs[(row0 + r) * D + d];
    sobs[r * lda_o + kperm(d, Dp)] = vv;
  }
  for (int ix = tid; ix < ROWS * H; ix += 256) {
    int r = ix / H, d = ix % H;
    float vv = 0.0f;
    int64_t row = row0 + r;
    if (row < p.E) {
      vv = p.hx[row * H + d];
      if (p.done_prev && p.done_prev[row]) vv = 0.0f;
    }
    shx[r * lda_h + kperm(d, H)] = vv;
  }
  __syncthreads();

  // input layer: x = tanh(obs @ Win + bin)
  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_core_16x64(sobs, lda_o, Dp, p.wtin, H, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        sx[(crow + i) * lda_h + kperm(col, H)] =
            tanhf(acc[n][i] + p.bin[col]);
      }
  }
  __syncthreads();

  // gates per 64-col chunk: 6 GEMM accumulations in registers
  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 gir[4] = {}, giz[4] = {}, gin[4] = {};
    f32x4 ghr[4] = {}, ghz[4] = {}, ghn[4] = {};
    mfma_core_16x64(sx, lda_h, H, p.wih, 3 * H, 0 * H + c0, lane, gir);
    mfma_core_16x64(sx, lda_h, H, p.wih, 3 * H, 1 * H + c0, lane, giz);
    mfma_core_16x64(sx, lda_h, H, p.wih, 3 * H, 2 * H + c0, lane, gin);
    mfma_core_16x64(shx, lda_h, H, p.whh, 3 * H, 0 * H + c0, lane, ghr);
    mfma_core_16x64(shx, lda_h, H, p.whh, 3 * H, 1 * H + c0, lane, ghz);
    mfma_core_16x64(shx, lda_h, H, p.whh, 3 * H, 2 * H + c0, lane, ghn);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        int r = crow + i;
        float rr = sigmoidf_(gir[n][i] + p.bih[col]
                             + ghr[n][i] + p.bhh[col]);
        float zz = sigmoidf_(giz[n][i] + p.bih[H + col]
                             + ghz[n][i] + p.bhh[H + col]);
        float nn = tanhf(gin[n][i] + p.bih[2 * H + col]
                         + rr * (ghn[n][i] + p.bhh[2 * H + col]));
        float hold = shx[r * lda_h + kperm(col, H)];
        float hnew = (1.0f - zz) * nn + zz * hold;
        shn[r * lda_h + kperm(col, H)] = hnew;
        int64_t row = row0 + r;
        if (row < p.E) p.hx_out[row * H + col] = hnew;
      }
  }
  __syncthreads();

  // head + sampling on h' (same as MLP actor)
  if (w == 0) {
    f32x4 acc[2] = {};
    const int kk = lane >> 4, bc = lane & 15;
    const float* arow = shn + (lane & 15) * lda_h + kk * (H >> 2);
    const float* bbase = p.wth + (int64_t)4 * (kk * HEADC + bc);
    for (int k16 = 0; k16 < (H >> 4); ++k16) {
      f32x4 av = *(const f32x4*)(arow + 4 * k16);
      const float* bk = bbase + (int64_t)16 * k16 * HEADC;
      #pragma unroll
      for (int n = 0; n < 2; ++n) {
        f32x4 bv = *(const f32x4*)(bk + 64 * n);
        #pragma unroll
        for (int s = 0; s < 4; ++s)
          acc[n] = __builtin_amdgcn_mfma_f32_16x16x4f32(av[s], bv[s], acc[n],
                                                        0, 0, 0);
      }
    }
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 2; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = n * 16 + ccol;
        shead[(crow + i) * (HEADC + 1) + col] = acc[n][i] + p.bh[col];
      }
  }
  __syncthreads();

  if (w == 0 && lane < ROWS) {
    int64_t row = row0 + lane;
    if (row < p.E) {
      const float* hrow = shead + lane * (HEADC + 1);
      p.value[row] = hrow[A];
      if (!p.value_only) {
        const uint8_t* mrow = p.mask + row * A;
        float mx = -3.0e38f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) mx = fmaxf(mx, hrow[i]);
        float sum = 0.0f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) sum += __expf(hrow[i] - mx);
        float thresh = p.u[row] * sum;
        float s = 0.0f;
        int act = -1;
        for (int i = 0; i < A; ++i) {
          if (!mrow[i]) continue;
          s += __expf(hrow[i] - mx);
          act = i;
          if (s >= thresh) break;
        }
        p.actions[row] = act;
        p.logp[row] = (hrow[act] - mx) - __logf(sum);
      }
    }
  }
}

}  // namespace

extern "C" int rlsched_gru_actor(const int64_t* ip, const uint64_t* bufs,
                                 void* stream) {
  GruActorArgs p;
  p.E = ip[0]; p.D = ip[1]; p.H = ip[2]; p.A = ip[3]; p.value_only = ip[4];
  if (p.H % 64 || p.A + 1 > HEADC) return -1;
  p.obs = (const float*)bufs[0];
  p.mask = (const uint8_t*)bufs[1];
  p.u = (const float*)bufs[2];
  p.done_prev = (const uint8_t*)bufs[3];   // 0 -> no reset
  p.hx = (const float*)bufs[4];
  p.hx_out = (float*)bufs[5];
  p.wtin = (const float*)bufs[6];
  p.bin = (const float*)bufs[7];
  p.wih = (const float*)bufs[8];
  p.bih = (const float*)bufs[9];
  p.whh = (const float*)bufs[10];
  p.bhh = (const float*)bufs[11];
  p.wth = (const float*)bufs[12];
  p.bh = (const float*)bufs[13];
  p.actions = (int64_t*)bufs[14];
  p.logp = (float*)bufs[15];
  p.value = (float*)bufs[16];
  const int Dp = ((int)p.D + 15) & ~15;
  size_t lds = sizeof(float) * (ROWS * (Dp + 4) + 3 * ROWS * ((int)p.H + 4)
                                + ROWS * (HEADC + 1));
  dim3 grid((unsigned)((p.E + ROWS - 1) / ROWS));
  hipLaunchKernelGGL(gru_actor_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

// ------------------------------------------------- bf16 fused MLP actor
// Same structure as actor_mlp_kernel but the GEMMs run on
// mfma_f32_16x16x32_bf16 (~13x the f32 MFMA rate; the actor is
// MFMA-issue-bound per PMC). Weights/activations are bf16, accumulation
// and the head/sampling stay f32. Fragment k-map: both operands use the
// CONTIGUOUS per-lane map k = kk*8 + j (any bijective map shared by A and
// B yields the correct dot product — verified by tools/probe_bf16_mfma).
// B packed by models/hip_actor.py: wt[K,N].view(K/32,4(kk),8(j),N)
// .permute(0,1,3,2) -> lane (kk,bc) reads ONE bf16x8 per 32-deep K step.
namespace {

typedef rls_bf16 bf16;

struct ActorBf16Args {
  int64_t E, D, H, A;
  int64_t value_only;
  const float* obs;
  const uint8_t* mask;
  const float* u;
  const bf16* wt1;      // packed [Dp,H] bf16
  const float* b1;
  const bf16* wt2;      // packed [H,H]
  const float* b2;
  const bf16* wth;      // packed [H,32]
  const float* bh;
  int64_t* actions;
  float* logp;
  float* value;
};

// C[16 x 64] for one wave: A bf16 LDS tile (PLAIN row-major, contiguous k),
// B packed global bf16. acc stays f32.
__device__ __forceinline__ void mfma_core_bf16_16x64(
    const bf16* __restrict__ a_lds, int lda, int K,
    const bf16* __restrict__ b_pk, int ldb, int col0, int lane,
    f32x4 acc[4]) {
  const int ar = lane & 15;
  const int kk = lane >> 4;
  const int bc = lane & 15;
  const bf16* arow = a_lds + ar * lda + kk * 8;
  const bf16* bbase = b_pk + (int64_t)8 * (kk * ldb + col0 + bc);
  for (int k32 = 0; k32 < (K >> 5); ++k32) {
    bf16x8 av = *(const bf16x8*)(arow + 32 * k32);
    const bf16* bk = bbase + (int64_t)32 * k32 * ldb;
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      bf16x8 bv = *(const bf16x8*)(bk + 128 * n);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc[n],
                                                       0, 0, 0);
    }
  }
}

__global__ __launch_bounds__(256) void actor_bf16_kernel(ActorBf16Args p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * ROWS;
  const int D = (int)p.D, H = (int)p.H, A = (int)p.A;
  const int Dp = (D + 31) & ~31;
  // bf16 row strides: even (16B-alignable) and stride/2 dwords != 0 mod 64
  const int lda_o = Dp + 8, lda_h = H + 8;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* sobs = (bf16*)smem;                    // [ROWS][Dp+8]
  bf16* h1 = sobs + ROWS * lda_o;              // [ROWS][H+8]
  bf16* h2 = h1 + ROWS * lda_h;                // [ROWS][H+8]
  float* shead = (float*)(h2 + ROWS * lda_h);  // [ROWS][33] f32

  for (int ix = tid; ix < ROWS * Dp; ix += 256) {
    int r = ix / Dp, d = ix % Dp;
    float vv = 0.0f;
    if (d < D && row0 + r < p.E) vv = p.obs[(row0 + r) * D + d];
    sobs[r * lda_o + d] = (bf16)vv;
  }
  __syncthreads();
  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_core_bf16_16x64(sobs, lda_o, Dp, p.wt1, H, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        h1[(crow + i) * lda_h + col] = (bf16)tanhf(acc[n][i] + p.b1[col]);
      }
  }
  __syncthreads();
  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_core_bf16_16x64(h1, lda_h, H, p.wt2, H, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        h2[(crow + i) * lda_h + col] = (bf16)tanhf(acc[n][i] + p.b2[col]);
      }
  }
  __syncthreads();
  if (w == 0) {
    f32x4 acc[2] = {};
    const int ar = lane & 15, kk = lane >> 4, bc = lane & 15;
    const bf16* arow = h2 + ar * lda_h + kk * 8;
    const bf16* bbase = p.wth + (int64_t)8 * (kk * HEADC + bc);
    for (int k32 = 0; k32 < (H >> 5); ++k32) {
      bf16x8 av = *(const bf16x8*)(arow + 32 * k32);
      const bf16* bk = bbase + (int64_t)32 * k32 * HEADC;
      #pragma unroll
      for (int n = 0; n < 2; ++n) {
        bf16x8 bv = *(const bf16x8*)(bk + 128 * n);
        acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc[n],
                                                         0, 0, 0);
      }
    }
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 2; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = n * 16 + ccol;
        shead[(crow + i) * (HEADC + 1) + col] = acc[n][i] + p.bh[col];
      }
  }
  __syncthreads();
  if (w == 0 && lane < ROWS) {
    int64_t row = row0 + lane;
    if (row < p.E) {
      const float* hrow = shead + lane * (HEADC + 1);
      p.value[row] = hrow[A];
      if (!p.value_only) {
        const uint8_t* mrow = p.mask + row * A;
        float mx = -3.0e38f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) mx = fmaxf(mx, hrow[i]);
        float sum = 0.0f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) sum += __expf(hrow[i] - mx);
        float thresh = p.u[row] * sum;
        float s = 0.0f;
        int act = -1;
        for (int i = 0; i < A; ++i) {
          if (!mrow[i]) continue;
          s += __expf(hrow[i] - mx);
          act = i;
          if (s >= thresh) break;
        }
        p.actions[row] = act;
        p.logp[row] = (hrow[act] - mx) - __logf(sum);
      }
    }
  }
}

}  // namespace

extern "C" int rlsched_actor_bf16(const int64_t* ip, const uint64_t* bufs,
                                  void* stream) {
  ActorBf16Args p;
  p.E = ip[0]; p.D = ip[1]; p.H = ip[2]; p.A = ip[3]; p.value_only = ip[4];
  if (p.H % 64 || p.H % 32 || p.A + 1 > HEADC) return -1;
  p.obs = (const float*)bufs[0];
  p.mask = (const uint8_t*)bufs[1];
  p.u = (const float*)bufs[2];
  p.wt1 = (const bf16*)bufs[3];
  p.b1 = (const float*)bufs[4];
  p.wt2 = (const bf16*)bufs[5];
  p.b2 = (const float*)bufs[6];
  p.wth = (const bf16*)bufs[7];
  p.bh = (const float*)bufs[8];
  p.actions = (int64_t*)bufs[9];
  p.logp = (float*)bufs[10];
  p.value = (float*)bufs[11];
  const int Dp = ((int)p.D + 31) & ~31;
  size_t lds = sizeof(bf16) * (ROWS * (Dp + 8) + 2 * ROWS * ((int)p.H + 8))
             + sizeof(float) * ROWS * (HEADC + 1);
  dim3 grid((unsigned)((p.E + ROWS - 1) / ROWS));
  hipLaunchKernelGGL(actor_bf16_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

// ------------------------------------------------- bf16 fused GRU actor
// bf16-MFMA variant of gru_actor_kernel (same contiguous shared k-map as
// the bf16 MLP actor; f32 accumulate, gates, head and sampling).
namespace {

struct GruBf16Args {
  int64_t E, D, H, A;
  int64_t value_only;
  const float* obs;
  const uint8_t* mask;
  const float* u;
  const uint8_t* done_prev;
  const float* hx;
  float* hx_out;
  const bf16* wtin;     // packed bf16 [Dp,H]
  const float* bin;
  const bf16* wih;      // packed [H,3H]
  const float* bih;
  const bf16* whh;      // packed [H,3H]
  const float* bhh;
  const bf16* wth;      // packed [H,32]
  const float* bh;
  int64_t* actions;
  float* logp;
  float* value;
};

__global__ __launch_bounds__(256) void gru_bf16_kernel(GruBf16Args p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * ROWS;
  const int D = (int)p.D, H = (int)p.H, A = (int)p.A;
  const int Dp = (D + 31) & ~31;
  const int lda_o = Dp + 8, lda_h = H + 8;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* sobs = (bf16*)smem;                  // [ROWS][Dp+8]
  bf16* sx = sobs + ROWS * lda_o;            // [ROWS][H+8]
  bf16* shx = sx + ROWS * lda_h;             // [ROWS][H+8]
  bf16* shn = shx + ROWS * lda_h;            // [ROWS][H+8] (h')
  float* shxf = (float*)(shn + ROWS * lda_h);  // [ROWS][H] f32 h~ (exact z*h)
  float* shead = shxf + ROWS * H;            // [ROWS][33]

  for (int ix = tid; ix < ROWS * Dp; ix += 256) {
    int r = ix / Dp, d = ix % Dp;
    float vv = 0.0f;
    if (d < D && row0 + r < p.E) vv = p.obs[(row0 + r) * D + d];
    sobs[r * lda_o + d] = (bf16)vv;
  }
  for (int ix = tid; ix < ROWS * H; ix += 256) {
    int r = ix / H, d = ix % H;
    float vv = 0.0f;
    int64_t row = row0 + r;
    if (row < p.E) {
      vv = p.hx[row * H + d];
      if (p.done_prev && p.done_prev[row]) vv = 0.0f;
    }
    shx[r * lda_h + d] = (bf16)vv;
    shxf[r * H + d] = vv;
  }
  __syncthreads();
  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_core_bf16_16x64(sobs, lda_o, Dp, p.wtin, H, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        sx[(crow + i) * lda_h + col] = (bf16)tanhf(acc[n][i] + p.bin[col]);
      }
  }
  __syncthreads();
  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 gir[4] = {}, giz[4] = {}, gin[4] = {};
    f32x4 ghr[4] = {}, ghz[4] = {}, ghn[4] = {};
    mfma_core_bf16_16x64(sx, lda_h, H, p.wih, 3 * H, 0 * H + c0, lane, gir);
    mfma_core_bf16_16x64(sx, lda_h, H, p.wih, 3 * H, 1 * H + c0, lane, giz);
    mfma_core_bf16_16x64(sx, lda_h, H, p.wih, 3 * H, 2 * H + c0, lane, gin);
    mfma_core_bf16_16x64(shx, lda_h, H, p.whh, 3 * H, 0 * H + c0, lane, ghr);
    mfma_core_bf16_16x64(shx, lda_h, H, p.whh, 3 * H, 1 * H + c0, lane, ghz);
    mfma_core_bf16_16x64(shx, lda_h, H, p.whh, 3 * H, 2 * H + c0, lane, ghn);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        int r = crow + i;
        float rr = sigmoidf_(gir[n][i] + p.bih[col]
                             + ghr[n][i] + p.bhh[col]);
        float zz = sigmoidf_(giz[n][i] + p.bih[H + col]
                             + ghz[n][i] + p.bhh[H + col]);
        float nn = tanhf(gin[n][i] + p.bih[2 * H + col]
                         + rr * (ghn[n][i] + p.bhh[2 * H + col]));
        float hold = shxf[r * H + col];   // f32 carry for the z*h path
        float hnew = (1.0f - zz) * nn + zz * hold;
        shn[r * lda_h + col] = (bf16)hnew;
        int64_t row = row0 + r;
        if (row < p.E) p.hx_out[row * H + col] = hnew;
      }
  }
  __syncthreads();
  if (w == 0) {
    f32x4 acc[2] = {};
    const int ar = lane & 15, kk = lane >> 4, bc = lane & 15;
    const bf16* arow = shn + ar * lda_h + kk * 8;
    const bf16* bbase = p.wth + (int64_t)8 * (kk * HEADC + bc);
    for (int k32 = 0; k32 < (H >> 5); ++k32) {
      bf16x8 av = *(const bf16x8*)(arow + 32 * k32);
      const bf16* bk = bbase + (int64_t)32 * k32 * HEADC;
      #pragma unroll
      for (int n = 0; n < 2; ++n) {
        bf16x8 bv = *(const bf16x8*)(bk + 128 * n);
        acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc[n],
                                                         0, 0, 0);
      }
    }
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 2; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = n * 16 + ccol;
        shead[(crow + i) * (HEADC + 1) + col] = acc[n][i] + p.bh[col];
      }
  }
  __syncthreads();
  if (w == 0 && lane < ROWS) {
    int64_t row = row0 + lane;
    if (row < p.E) {
      const float* hrow = shead + lane * (HEADC + 1);
      p.value[row] = hrow[A];
      if (!p.value_only) {
        const uint8_t* mrow = p.mask + row * A;
        float mx = -3.0e38f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) mx = fmaxf(mx, hrow[i]);
        float sum = 0.0f;
        for (int i = 0; i < A; ++i)
          if (mrow[i]) sum += __expf(hrow[i] - mx);
        float thresh = p.u[row] * sum;
        float s = 0.0f;
        int act = -1;
        for (int i = 0; i < A; ++i) {
          if (!mrow[i]) continue;
          s += __expf(hrow[i] - mx);
          act = i;
          if (s >= thresh) break;
        }
        p.actions[row] = act;
        p.logp[row] = (hrow[act] - mx) - __logf(sum);
      }
    }
  }
}

}  // namespace

extern "C" int rlsched_gru_bf16(const int64_t* ip, const uint64_t* bufs,
                                void* stream) {
  GruBf16Args p;
  p.E = ip[0]; p.D = ip[1]; p.H = ip[2]; p.A = ip[3]; p.value_only = ip[4];
  if (p.H % 64 || p.A + 1 > HEADC) return -1;
  p.obs = (const float*)bufs[0];
  p.mask = (const uint8_t*)bufs[1];
  p.u = (const float*)bufs[2];
  p.done_prev = (const uint8_t*)bufs[3];
  p.hx = (const float*)bufs[4];
  p.hx_out = (float*)bufs[5];
  p.wtin = (const bf16*)bufs[6];
  p.bin = (const float*)bufs[7];
  p.wih = (const bf16*)bufs[8];
  p.bih = (const float*)bufs[9];
  p.whh = (const bf16*)bufs[10];
  p.bhh = (const float*)bufs[11];
  p.wth = (const bf16*)bufs[12];
  p.bh = (const float*)bufs[13];
  p.actions = (int64_t*)bufs[14];
  p.logp = (float*)bufs[15];
  p.value = (float*)bufs[16];
  const int Dp = ((int)p.D + 31) & ~31;
  size_t lds = sizeof(bf16) * (ROWS * (Dp + 8) + 3 * ROWS * ((int)p.H + 8))
             + sizeof(float) * (ROWS * (int)p.H + ROWS * (HEADC + 1));
  dim3 grid((unsigned)((p.E + ROWS - 1) / ROWS));
  hipLaunchKernelGGL(gru_bf16_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

// Build provenance: ops/build.py embeds a sha256-prefix of this lib's
// sources at compile time and verifies it at load (stale-binary guard,
// VERDICT r1 weak #6). The "RLSHASH:" marker makes the hash findable by
// byte-scan without dlopen.
#ifndef RLS_SOURCE_HASH
#define RLS_SOURCE_HASH "unset"
#endif
extern "C" const char* rlsched_source_hash() {
  return "RLSHASH:" RLS_SOURCE_HASH;
}
