// Shared DEVICE-side implementation of the batched DES cluster env.
// Single source of truth for the transition semantics: included by BOTH
// env_step.hip (the standalone per-step kernel) and rollout_mega.hip
// (the fused whole-rollout kernel), so CPU-oracle parity proven for one
// is parity for both. Anonymous namespace: every TU gets its own inline
// copies (no ODR issues; all functions are __device__ __forceinline__ or
// internal).
#pragma once
#include "env_common.h"

namespace {


__device__ __forceinline__ int64_t wave_min_i64(int64_t v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    int64_t o = __shfl_xor(v, off, RLS_WAVE);
    v = o < v ? o : v;
  }
  return v;
}

__device__ __forceinline__ int wave_sum_i32(int v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, RLS_WAVE);
  return v;
}

__device__ __forceinline__ int wave_max_i32(int v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = max(v, __shfl_xor(v, off, RLS_WAVE));
  return v;
}

// float32 math mirrored by cpu_ref._lognorm
__device__ __forceinline__ float lognorm(int64_t x_us) {
  return log1pf((float)x_us / 1e6f) / 10.0f;
}

// Register-resident per-env scalars (written back once per kernel).
struct Sc {
  int64_t clock, cursor, jct_sum;
  int pend_count, run_count, decisions, completed;
  float reward_sum;
};

// Per-env view: every pointer offset to env e.
struct EnvView {
  const RlsParams* p;
  const int64_t *trace_arr, *trace_work;
  const int32_t *trace_gpus, *trace_ming;
  int64_t job_lo, job_hi, base;
  int32_t *node_free;        // LDS slice during kernel execution
  int32_t *node_free_glb;    // global backing (written back at the end)
  int32_t *run_valid_glb;    // global backing when run state is in LDS
  int64_t *run_finish_glb;   //   (run_state_to_lds/run_state_writeback)
  int32_t *pend_job;
  int64_t *pend_override;
  int32_t *run_valid, *run_job, *run_gpus;
  int64_t *run_finish, *run_start;
  int32_t *run_alloc;
  float *obs;
  uint8_t *mask;
};

__device__ EnvView make_view(const RlsParams& p, const RlsBufs& b, int64_t e) {
  EnvView v;
  v.p = &p;
  v.trace_arr = b.trace_arr; v.trace_gpus = b.trace_gpus; v.trace_work = b.trace_work;
  v.trace_ming = b.trace_ming;
  v.job_lo = b.job_lo[e]; v.job_hi = b.job_hi[e]; v.base = b.base[e];
  v.node_free = b.node_free + e * p.N;
  v.pend_job = b.pend_job + e * p.P;
  v.pend_override = b.pend_override + e * p.P;
  v.run_valid = b.run_valid + e * p.R;
  v.run_job = b.run_job + e * p.R;
  v.run_gpus = b.run_gpus + e * p.R;
  v.run_finish = b.run_finish + e * p.R;
  v.run_start = b.run_start + e * p.R;
  v.run_alloc = b.run_alloc + e * p.R * p.S;
  v.obs = b.obs + e * p.obs_dim;
  v.mask = b.mask + e * p.A;
  return v;
}

__device__ __forceinline__ void load_sc(const RlsBufs& b, int64_t e, Sc& s) {
  s.clock = b.clock_[e]; s.cursor = b.cursor[e]; s.jct_sum = b.jct_sum[e];
  s.pend_count = b.pend_count[e]; s.run_count = b.run_count[e];
  s.decisions = b.decisions[e]; s.completed = b.completed[e];
  s.reward_sum = b.reward_sum[e];
}

__device__ __forceinline__ void store_sc(const RlsBufs& b, int64_t e, const Sc& s) {
  b.clock_[e] = s.clock; b.cursor[e] = s.cursor; b.jct_sum[e] = s.jct_sum;
  b.pend_count[e] = s.pend_count; b.run_count[e] = s.run_count;
  b.decisions[e] = s.decisions; b.completed[e] = s.completed;
  b.reward_sum[e] = s.reward_sum;
}

__device__ __forceinline__ int64_t arrival_of(const EnvView& v, int64_t j) {
  return v.trace_arr[j] - v.base;
}

// Deterministic first-fit (ENV_SPEC): scan nodes in index order. Returns
// span count, or -1 on failure. pairs filled when non-null. Runs redundantly
// on all lanes (or divergently per-lane for the obs fits bits).
__device__ int first_fit(const EnvView& v, int g, int* pairs_n, int* pairs_c) {
  int left = g, cnt = 0;
  const int N = (int)v.p->N, S = (int)v.p->S;
  for (int n = 0; n < N; ++n) {
    int f = v.node_free[n];
    if (f > 0) {
      if (cnt == S) return -1;
      int take = f < left ? f : left;
      if (pairs_n) { pairs_n[cnt] = n; pairs_c[cnt] = take; }
      ++cnt;
      left -= take;
      if (left == 0) return cnt;
    }
  }
  return -1;
}

// Next event time (RLS_NO_EVENT if none). Wave-parallel over run slots.
__device__ int64_t next_event(const EnvView& v, const Sc& s, int lane) {
  int64_t m = RLS_NO_EVENT;
  const int R = (int)v.p->R;
  for (int r = lane; r < R; r += RLS_WAVE) {
    if (v.run_valid[r]) {
      int64_t f = v.run_finish[r];
      m = f < m ? f : m;
    }
  }
  m = wave_min_i64(m);
  if (s.cursor < v.job_hi) {
    int64_t a = arrival_of(v, s.cursor);
    m = a < m ? a : m;
  }
  return m;
}

// Admission (redundant-scalar).
__device__ void admit(const EnvView& v, Sc& s) {
  const int P = (int)v.p->P;
  while (s.cursor < v.job_hi && arrival_of(v, s.cursor) <= s.clock
         && s.pend_count < P) {
    v.pend_job[s.pend_count] = (int32_t)s.cursor;
    v.pend_override[s.pend_count] = 0;
    ++s.pend_count; ++s.cursor;
  }
}

// Remaining work of pending position pos, in GPU-us (ENV_SPEC v3;
// pend_override stores GPU-us; exact for rigid jobs).
__device__ __forceinline__ int64_t pend_rem_gpuus(const EnvView& v, int pos) {
  int64_t o = v.pend_override[pos];
  if (o > 0) return o;
  int32_t j = v.pend_job[pos];
  return v.trace_work[j] * (int64_t)v.trace_gpus[j];
}

// Obs work feature: remaining wall-time at max demand.
__device__ __forceinline__ int64_t pend_work_us(const EnvView& v, int pos) {
  int64_t o = v.pend_override[pos];
  int32_t j = v.pend_job[pos];
  if (o > 0) {
    int64_t g = v.trace_gpus[j];
    return (o + g - 1) / g;
  }
  return v.trace_work[j];
}

// Descending-g placement scan. Returns allocated g (pairs filled) or -1.
__device__ int try_fit(const EnvView& v, int32_t j, int* pn, int* pc_,
                       int* span_out) {
  int g_max = v.trace_gpus[j];
  int g_min = v.p->elastic ? v.trace_ming[j] : g_max;
  for (int g = g_max; g >= g_min; --g) {
    int span = first_fit(v, g, pn, pc_);
    if (span >= 0) { *span_out = span; return g; }
  }
  return -1;
}

// Remove pending position a: wave-parallel left shift. Cross-chunk
// hazard-free: chunk c+1's reads start past chunk c's last write, and within
// a chunk the wave's loads complete before its stores issue (lockstep).
__device__ void pend_remove(const EnvView& v, Sc& s, int a, int lane) {
  int pc = s.pend_count;
  for (int basep = a; basep < pc - 1; basep += RLS_WAVE) {
    int idx = basep + lane;
    int32_t j = 0; int64_t o = 0;
    bool act = idx < pc - 1;
    if (act) { j = v.pend_job[idx + 1]; o = v.pend_override[idx + 1]; }
    if (act) { v.pend_job[idx] = j; v.pend_override[idx] = o; }
  }
  s.pend_count = pc - 1;
}

// Lowest invalid running slot (redundant-scalar; R <= 512 so cheap scan).
__device__ int free_slot(const EnvView& v) {
  const int R = (int)v.p->R;
  for (int r = 0; r < R; ++r)
    if (!v.run_valid[r]) return r;
  return -1;  // unreachable: R >= max concurrent jobs by construction
}

// Schedule pending position a. Returns true on success.
__device__ bool place(const EnvView& v, Sc& s, int a, int lane) {
  int32_t j = v.pend_job[a];
  int pn[RLS_MAX_SPAN], pc_[RLS_MAX_SPAN];
  int span;
  int g = try_fit(v, j, pn, pc_, &span);
  if (g < 0) return false;
  int slot = free_slot(v);
  if (slot < 0) return false;
  for (int i = 0; i < span; ++i) v.node_free[pn[i]] -= pc_[i];
  int64_t rem = pend_rem_gpuus(v, a);
  v.run_valid[slot] = 1;
  v.run_job[slot] = j;
  v.run_gpus[slot] = g;   // ALLOCATED gpus (== demand when rigid)
  v.run_start[slot] = s.clock;
  v.run_finish[slot] = s.clock + (rem + g - 1) / g;
  const int S = (int)v.p->S;
  for (int i = 0; i < S; ++i)
    v.run_alloc[slot * S + i] = i < span ? ((pn[i] << 16) | pc_[i]) : 0;
  s.run_count += 1;
  pend_remove(v, s, a, lane);
  return true;
}

// Release slot r's GPUs (redundant-scalar).
__device__ void release_slot(const EnvView& v, Sc& s, int r) {
  const int S = (int)v.p->S;
  for (int i = 0; i < S; ++i) {
    int32_t pk = v.run_alloc[r * S + i];
    int cnt = pk & 0xFFFF;
    if (cnt > 0) v.node_free[pk >> 16] += cnt;
  }
  v.run_valid[r] = 0;
  s.run_count -= 1;
}

// NOOP transition: advance time, complete, admit. Returns reward.
__device__ float advance(const EnvView& v, Sc& s, int lane) {
  int64_t t_next = next_event(v, s, lane);
  if (t_next == RLS_NO_EVENT) t_next = s.clock + v.p->idle_tick_us;
  int64_t dt = t_next - s.clock;
  int n_sys = s.pend_count + s.run_count;
  float reward = -((float)n_sys * (float)dt) / (float)v.p->time_scale_us;
  s.clock = t_next;
  // completions, ballot-scanned in slot order (sums are commutative but
  // keep the order deterministic anyway)
  const int R = (int)v.p->R;
  for (int cb = 0; cb < R; cb += RLS_WAVE) {
    int r = cb + lane;
    bool comp = (r < R) && v.run_valid[r] && v.run_finish[r] <= s.clock;
    uint64_t m = __ballot(comp);
    while (m) {
      int bbit = __ffsll((unsigned long long)m) - 1;
      m &= m - 1;
      int r2 = cb + bbit;
      int32_t j = v.run_job[r2];
      s.completed += 1;
      s.jct_sum += s.clock - arrival_of(v, j);
      release_slot(v, s, r2);
    }
  }
  admit(v, s);
  return reward;
}

// Gandiva MIGRATE (ENV_SPEC v2): re-place widest-span running job.
__device__ bool do_migrate(const EnvView& v, Sc& s) {
  if (s.run_count == 0) return false;
  const int R = (int)v.p->R, S = (int)v.p->S;
  int best_r = -1, best_span = -1;
  for (int r = 0; r < R; ++r) {
    if (!v.run_valid[r]) continue;
    int span = 0;
    for (int i = 0; i < S; ++i) span += (v.run_alloc[r * S + i] & 0xFFFF) > 0;
    if (span > best_span) { best_span = span; best_r = r; }
  }
  int old_pk[RLS_MAX_SPAN];
  for (int i = 0; i < S; ++i) {
    old_pk[i] = v.run_alloc[best_r * S + i];
    int cnt = old_pk[i] & 0xFFFF;
    if (cnt > 0) v.node_free[old_pk[i] >> 16] += cnt;
  }
  int pn[RLS_MAX_SPAN], pc_[RLS_MAX_SPAN];
  int span;
  int g_old = v.run_gpus[best_r];
  int g_new = try_fit(v, v.run_job[best_r], pn, pc_, &span);
  if (g_new < 0) {  // restore, degrade to NOOP
    for (int i = 0; i < S; ++i) {
      int cnt = old_pk[i] & 0xFFFF;
      if (cnt > 0) v.node_free[old_pk[i] >> 16] -= cnt;
    }
    return false;
  }
  for (int i = 0; i < span; ++i) v.node_free[pn[i]] -= pc_[i];
  for (int i = 0; i < S; ++i)
    v.run_alloc[best_r * S + i] = i < span ? ((pn[i] << 16) | pc_[i]) : 0;
  if (g_new != g_old) {  // elastic re-placement changes speed
    int64_t rem0 = s.clock < v.run_finish[best_r]
                   ? v.run_finish[best_r] - s.clock : 1;
    int64_t rem = rem0 * g_old;
    v.run_gpus[best_r] = g_new;
    v.run_finish[best_r] = s.clock + (rem + g_new - 1) / g_new;
  }
  return true;
}

// Gandiva TIMESLICE: preempt longest-elapsed running job to pending tail.
__device__ bool do_timeslice(const EnvView& v, Sc& s) {
  if (s.run_count == 0 || s.pend_count >= (int)v.p->P) return false;
  const int R = (int)v.p->R;
  int best_r = -1; int64_t best_el = -1;
  for (int r = 0; r < R; ++r) {
    if (!v.run_valid[r]) continue;
    int64_t el = s.clock - v.run_start[r];
    if (el > best_el) { best_el = el; best_r = r; }
  }
  int64_t remaining = v.run_finish[best_r] - s.clock;
  if (remaining < 1) remaining = 1;
  remaining *= v.run_gpus[best_r];   // GPU-us (ENV_SPEC v3)
  v.pend_job[s.pend_count] = v.run_job[best_r];
  v.pend_override[s.pend_count] = remaining;
  s.pend_count += 1;
  release_slot(v, s, best_r);
  return true;
}

// ENV_SPEC v5 SHRINK: halve the widest running elastic job (largest g
// with g/2 >= min_gpus; tie -> lowest slot). Redundant-scalar.
__device__ bool do_shrink(const EnvView& v, Sc& s) {
  const int R = (int)v.p->R, S = (int)v.p->S;
  int best_g = -1, slot = -1;
  for (int r = 0; r < R; ++r) {
    if (!v.run_valid[r]) continue;
    int g = v.run_gpus[r];
    int32_t j = v.run_job[r];
    if (g >= 2 && g / 2 >= v.trace_ming[j] && g > best_g) {
      best_g = g; slot = r;
    }
  }
  if (slot < 0) return false;
  int g_old = best_g, g_new = g_old / 2;
  int64_t rem0 = s.clock < v.run_finish[slot]
                 ? v.run_finish[slot] - s.clock : 1;
  int64_t rem = rem0 * g_old;
  int old_pk[RLS_MAX_SPAN];
  for (int i = 0; i < S; ++i) {
    old_pk[i] = v.run_alloc[slot * S + i];
    int cnt = old_pk[i] & 0xFFFF;
    if (cnt > 0) v.node_free[old_pk[i] >> 16] += cnt;
  }
  int pn[RLS_MAX_SPAN], pc_[RLS_MAX_SPAN];
  int span = first_fit(v, g_new, pn, pc_);
  if (span < 0) {  // span-constrained fit can fail: restore, NOOP
    for (int i = 0; i < S; ++i) {
      int cnt = old_pk[i] & 0xFFFF;
      if (cnt > 0) v.node_free[old_pk[i] >> 16] -= cnt;
    }
    return false;
  }
  for (int i = 0; i < span; ++i) v.node_free[pn[i]] -= pc_[i];
  for (int i = 0; i < S; ++i)
    v.run_alloc[slot * S + i] = i < span ? ((pn[i] << 16) | pc_[i]) : 0;
  v.run_gpus[slot] = g_new;
  v.run_finish[slot] = s.clock + (rem + g_new - 1) / g_new;
  return true;
}

// ENV_SPEC v5 EXPAND: grow the most-shrunk running job (largest demand
// deficit; tie -> lowest slot) to the largest g' in (g, min(demand, 2g)].
__device__ bool do_expand(const EnvView& v, Sc& s) {
  const int R = (int)v.p->R, S = (int)v.p->S;
  int best_d = 0, slot = -1;
  for (int r = 0; r < R; ++r) {
    if (!v.run_valid[r]) continue;
    int d = v.trace_gpus[v.run_job[r]] - v.run_gpus[r];
    if (d > best_d) { best_d = d; slot = r; }
  }
  if (slot < 0) return false;
  int g_old = v.run_gpus[slot];
  int32_t j = v.run_job[slot];
  int g_target = min((int)v.trace_gpus[j], 2 * g_old);
  int64_t rem0 = s.clock < v.run_finish[slot]
                 ? v.run_finish[slot] - s.clock : 1;
  int64_t rem = rem0 * g_old;
  int old_pk[RLS_MAX_SPAN];
  for (int i = 0; i < S; ++i) {
    old_pk[i] = v.run_alloc[slot * S + i];
    int cnt = old_pk[i] & 0xFFFF;
    if (cnt > 0) v.node_free[old_pk[i] >> 16] += cnt;
  }
  int pn[RLS_MAX_SPAN], pc_[RLS_MAX_SPAN];
  int g_new = -1, span = -1;
  for (int gt = g_target; gt > g_old; --gt) {
    span = first_fit(v, gt, pn, pc_);
    if (span >= 0) { g_new = gt; break; }
  }
  if (g_new < 0) {
    for (int i = 0; i < S; ++i) {
      int cnt = old_pk[i] & 0xFFFF;
      if (cnt > 0) v.node_free[old_pk[i] >> 16] -= cnt;
    }
    return false;
  }
  for (int i = 0; i < span; ++i) v.node_free[pn[i]] -= pc_[i];
  for (int i = 0; i < S; ++i)
    v.run_alloc[slot * S + i] = i < span ? ((pn[i] << 16) | pc_[i]) : 0;
  v.run_gpus[slot] = g_new;
  v.run_finish[slot] = s.clock + (rem + g_new - 1) / g_new;
  return true;
}

// Observation + action mask (ENV_SPEC layout). Wave-parallel over slots.
__device__ void write_obs_mask(const EnvView& v, const Sc& s, int lane) {
  const int K = (int)v.p->K, N = (int)v.p->N;
  const int OD = (int)v.p->obs_dim;
  const int64_t skip = v.p->debug_skip;
  if (skip & 2) return;   // ablation only (bench/abl_env.py)
  for (int i = lane; i < OD; i += RLS_WAVE) v.obs[i] = 0.0f;
  int ft = 0;
  for (int n = lane; n < N; n += RLS_WAVE) ft += v.node_free[n];
  ft = wave_sum_i32(ft);
  int64_t t_next = (skip & 4) ? 0 : next_event(v, s, lane);
  if (lane == 0) {
    v.obs[0] = (float)ft / (float)(v.p->N * v.p->G);
    v.obs[1] = (float)s.pend_count / (float)v.p->P;
    v.obs[2] = (float)s.run_count / (float)v.p->R;
    v.obs[3] = t_next != RLS_NO_EVENT ? lognorm(t_next - s.clock) : 0.0f;
  }
  // S-prefix free capacity: first_fit(g) succeeds iff g <= cap_S, the
  // free total of the FIRST max_span nonzero-free nodes in index order
  // (first-fit always consumes exactly that prefix — cpu_ref.fits_cap).
  // One redundant-scalar O(N) pass replaces K per-slot first-fit scans.
  int capS = 0;
  {
    int used = 0;
    const int S = (int)v.p->S;
    for (int n = 0; n < N && used < S; ++n) {
      int f = v.node_free[n];
      if (f > 0) { capS += f; ++used; }
    }
  }
  // per-slot features + mask: lane k handles slot k
  if (lane < K && !(skip & 1)) {
    bool valid = lane < s.pend_count;
    bool fits = false;
    if (valid) {
      int32_t j = v.pend_job[lane];
      int g = v.trace_gpus[j];
      int g_fit = v.p->elastic ? v.trace_ming[j] : g;
      fits = g_fit <= capS;
      int bix = 4 + 5 * lane;
      v.obs[bix + 0] = 1.0f;
      v.obs[bix + 1] = (float)g / (float)(v.p->N * v.p->G);
      v.obs[bix + 2] = lognorm(s.clock - arrival_of(v, j));
      v.obs[bix + 3] = lognorm(pend_work_us(v, lane));
      v.obs[bix + 4] = fits ? 1.0f : 0.0f;
    }
    v.mask[lane] = (uint8_t)(valid && fits);
  }
  // obs v2 extras (ENV_SPEC v4): written after the base layout so the
  // default obs_version=1 path is bit-identical to the original kernel
  if (v.p->obs_version >= 2) {
    const int EB = 4 + 5 * K, G = (int)v.p->G;
    int cnt[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    int mxf = 0;
    for (int n = lane; n < N; n += RLS_WAVE) {
      int f = v.node_free[n];
      int idx = (f * 8) / G;
      if (idx > 7) idx = 7;
      mxf = max(mxf, f);
      #pragma unroll
      for (int b = 0; b < 8; ++b) cnt[b] += (idx == b) ? 1 : 0;
    }
    mxf = wave_max_i32(mxf);
    #pragma unroll
    for (int b = 0; b < 8; ++b) {
      int tot = wave_sum_i32(cnt[b]);
      if (lane == 0) v.obs[EB + b] = (float)tot / (float)N;
    }
    if (lane == 0) {
      int over = s.pend_count - K;
      v.obs[EB + 8] = (over > 0 ? (float)over : 0.0f) / (float)v.p->P;
      v.obs[EB + 9] = (float)mxf / (float)G;
      int64_t span = v.job_hi - v.job_lo;
      if (span < 1) span = 1;
      v.obs[EB + 10] = (float)(s.cursor - v.job_lo) / (float)span;
      v.obs[EB + 11] = (float)s.decisions / (float)v.p->horizon;
    }
  }
  if (lane == (int)(K & 63)) v.mask[K] = 1;
  if (v.p->gandiva && lane == 0) {
    v.mask[K + 1] = (uint8_t)(s.run_count > 0);
    v.mask[K + 2] = (uint8_t)(s.run_count > 0 && s.pend_count < (int)v.p->P);
  }
  if (v.p->elastic_resize) {   // v5: wave-parallel candidate scan
    const int R = (int)v.p->R;
    bool can_s = false, can_e = false;
    for (int r = lane; r < R; r += RLS_WAVE) {
      if (!v.run_valid[r]) continue;
      int g = v.run_gpus[r];
      int32_t j = v.run_job[r];
      can_s |= (g >= 2 && g / 2 >= v.trace_ming[j]);
      can_e |= (g < v.trace_gpus[j]);
    }
    uint64_t bs = __ballot(can_s), be = __ballot(can_e);
    if (lane == 0) {
      int base = K + 1 + (v.p->gandiva ? 2 : 0);
      v.mask[base] = (uint8_t)(bs != 0);
      v.mask[base + 1] = (uint8_t)(be != 0);
    }
  }
}

// Fresh-episode state (parallel fills + redundant-scalar admit).
__device__ void reset_env(const EnvView& v, Sc& s, int lane) {
  const int N = (int)v.p->N, R = (int)v.p->R;
  for (int n = lane; n < N; n += RLS_WAVE) v.node_free[n] = (int32_t)v.p->G;
  for (int r = lane; r < R; r += RLS_WAVE) v.run_valid[r] = 0;
  s.clock = 0;
  s.cursor = v.job_lo;
  s.pend_count = 0;
  s.run_count = 0;
  s.decisions = 0;
  s.completed = 0;
  s.jct_sum = 0;
  s.reward_sum = 0.0f;
  admit(v, s);
}

// node_free lives in LDS for the kernel's duration: the first-fit scans
// (placement + K per-slot fits bits every step) are the hottest dependent
// loads in the transition; LDS turns them from L2 round trips into
// near-register accesses. Copied in at launch, written back at the end.
// (Each wave touches only its own slice — no barriers needed.)
#define RLS_MAX_NODES 128

__device__ __forceinline__ void nf_to_lds(EnvView& v, int32_t* s_nf,
                                          int lane) {
  const int N = (int)v.p->N;
  for (int n = lane; n < N; n += RLS_WAVE) s_nf[n] = v.node_free[n];
  v.node_free_glb = v.node_free;
  v.node_free = s_nf;
}

__device__ __forceinline__ void nf_writeback(const EnvView& v, int lane) {
  const int N = (int)v.p->N;
  for (int n = lane; n < N; n += RLS_WAVE) v.node_free_glb[n] = v.node_free[n];
}

// run_valid/run_finish LDS residency (round 2): the two next_event scans
// + the completion ballot re-read these R-slot arrays from L2 every
// step — the hottest dependent loads left after node_free moved to LDS
// (env_step was 27.8% of device time at the r1 headline). Same redirect
// pattern as nf_to_lds; all writers go through the EnvView pointers, so
// semantics are untouched. Wave-private slices, no barriers.
__device__ __forceinline__ void run_state_to_lds(EnvView& v, int32_t* s_rv,
                                                 int64_t* s_rf, int lane) {
  const int R = (int)v.p->R;
  for (int r = lane; r < R; r += RLS_WAVE) {
    s_rv[r] = v.run_valid[r];
    s_rf[r] = v.run_finish[r];
  }
  v.run_valid_glb = v.run_valid;
  v.run_finish_glb = v.run_finish;
  v.run_valid = s_rv;
  v.run_finish = s_rf;
}

__device__ __forceinline__ void run_state_writeback(const EnvView& v,
                                                    int lane) {
  const int R = (int)v.p->R;
  for (int r = lane; r < R; r += RLS_WAVE) {
    v.run_valid_glb[r] = v.run_valid[r];
    v.run_finish_glb[r] = v.run_finish[r];
  }
}

void unpack(const int64_t* params, const uint64_t* bufs, RlsParams& p, RlsBufs& b) {
  p.E = params[0]; p.N = params[1]; p.G = params[2]; p.K = params[3];
  p.P = params[4]; p.R = params[5]; p.S = params[6]; p.A = params[7];
  p.obs_dim = params[8]; p.idle_tick_us = params[9]; p.time_scale_us = params[10];
  p.horizon = params[11]; p.gandiva = params[12]; p.elastic = params[13];
  p.debug_skip = params[14];
  p.obs_version = params[15];
  p.elastic_resize = params[16];
  uint64_t* dst = reinterpret_cast<uint64_t*>(&b);
  for (int i = 0; i < RLS_NUM_BUFS; ++i) dst[i] = bufs[i];
}


}  // namespace
