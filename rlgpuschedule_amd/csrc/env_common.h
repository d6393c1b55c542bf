// Shared declarations for the batched on-device cluster-scheduling env.
// Semantics contract: docs/ENV_SPEC.md (bit-parity with env/cpu_ref.py).
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define RLS_WAVE 64
#define RLS_ENVS_PER_BLOCK 4
#define RLS_BLOCK (RLS_WAVE * RLS_ENVS_PER_BLOCK)
#define RLS_NO_EVENT 0x7FFFFFFFFFFFFFFFLL
#define RLS_MAX_SPAN 8  // compile-time cap on EnvSpec.max_span

struct RlsParams {
  int64_t E;          // number of envs
  int64_t N;          // nodes
  int64_t G;          // gpus per node
  int64_t K;          // visible queue slots
  int64_t P;          // pending capacity
  int64_t R;          // running slots
  int64_t S;          // max span (<= RLS_MAX_SPAN)
  int64_t A;          // num actions (K+1, +2 if gandiva)
  int64_t obs_dim;    // 4 + 5K
  int64_t idle_tick_us;
  int64_t time_scale_us;
  int64_t horizon;
  int64_t gandiva;    // 0/1
  int64_t elastic;    // 0/1 (ENV_SPEC v3)
  int64_t debug_skip; // ablation bitmask (bench only): 1=slot obs/fits,
                      // 2=whole obs/mask phase, 4=obs[3] next_event
  int64_t obs_version; // 1 = base layout; 2 adds 12 global features (v4)
  int64_t elastic_resize; // 0/1: SHRINK/EXPAND running jobs (v5)
};

// Device buffers; all row-major with env as the leading dim.
// Bound from Python as a uint64 pointer array in EXACTLY this field order —
// keep in sync with rlgpuschedule_amd/ops/envstep.py:BUF_ORDER.
struct RlsBufs {
  const int64_t* trace_arr;      // [J]
  const int32_t* trace_gpus;     // [J]
  const int64_t* trace_work;     // [J]
  const int32_t* trace_ming;     // [J] min demand (== gpus when rigid)
  const int64_t* job_lo;         // [E]
  const int64_t* job_hi;         // [E]
  const int64_t* base;           // [E]
  int64_t* clock_;               // [E]
  int64_t* cursor;               // [E]
  int32_t* node_free;            // [E*N]
  int32_t* pend_job;             // [E*P]
  int64_t* pend_override;        // [E*P]
  int32_t* pend_count;           // [E]
  int32_t* run_valid;            // [E*R]
  int32_t* run_job;              // [E*R]
  int32_t* run_gpus;             // [E*R]
  int64_t* run_finish;           // [E*R]
  int64_t* run_start;            // [E*R]
  int32_t* run_alloc;            // [E*R*S] packed (node<<16)|cnt
  int32_t* run_count;            // [E]
  int32_t* decisions;            // [E]
  int32_t* completed;            // [E]
  int64_t* jct_sum;              // [E]
  float*   reward_sum;           // [E]
  const int64_t* actions;        // [E] (step input)
  float*   obs;                  // [E*obs_dim] (output)
  float*   rewards;              // [E] (output)
  uint8_t* done;                 // [E] (output)
  uint8_t* mask;                 // [E*A] (output)
  int32_t* ep_count;             // [E] finished-episode accumulators
  float*   ep_return;            // [E]
  int64_t* ep_jct_sum;           // [E]
  int64_t* ep_completed;         // [E]
  int64_t* ep_decisions;         // [E]
};

#define RLS_NUM_BUFS 34

extern "C" {
// Both return 0 on success, a hipError_t otherwise. `stream` is a
// hipStream_t (pass torch.cuda.current_stream().cuda_stream).
int rlsched_env_reset(const int64_t* params, const uint64_t* bufs, void* stream);
int rlsched_env_step(const int64_t* params, const uint64_t* bufs, void* stream);
int rlsched_device_sync();
}
