// Hand-written CDNA4 training kernels for the PPO/IMPALA update path:
// MLP forward (activation-saving), masked PPO loss head (fwd+bwd), backward
// dX GEMMs with fused tanh', backward dW GEMMs (LDS-tiled, split over
// batch chunks with atomic accumulation), and bias column sums.
// All GEMMs are MFMA f32 16x16x4 (exact f32) with the kperm/f32x4 fragment
// scheme of nn_common.h. Replaces torch/hipBLASLt in the update so the
// whole PPO iteration is hand-written HIP (BASELINE.json north star).
//
// Gradient-check tests: tests/test_train_kernels.py vs torch autograd.

#include "nn_common.h"

#define ROWS 16
#define HEADC 32

namespace {

// --------------------------------------------------- training MLP forward
// Saves h1,h2 (plain [B,H]) and head [B,HEADC] for the backward pass.
struct FwdTrainArgs {
  int64_t B, D, H;
  const float* obs;   // [B,D]
  const float* wt1;   // packed [Dp,H]
  const float* b1;
  const float* wt2;   // packed [H,H]
  const float* b2;
  const float* wth;   // packed [H,HEADC]
  const float* bh;
  float* h1;          // [B,H] out (plain)
  float* h2;          // [B,H] out
  float* head;        // [B,HEADC] out
};

// gemm chunk with epilogue writing BOTH kperm LDS (next layer's A) and a
// plain global activation row block.
__device__ __forceinline__ void chunk_fwd_train(
    const float* a_lds, int lda, int K, const float* b_pk, int ldb, int col0,
    const float* bias, float* out_lds, int ldo, float* gout, int64_t row0,
    int64_t B, int ldg, int lane) {
  f32x4 acc[4] = {};
  mfma_core_16x64(a_lds, lda, K, b_pk, ldb, col0, lane, acc);
  const int crow = (lane >> 4) * 4, ccol = lane & 15;
  #pragma unroll
  for (int n = 0; n < 4; ++n)
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int col = col0 + n * 16 + ccol;
      float vv = tanhf(acc[n][i] + bias[col]);
      out_lds[(crow + i) * ldo + kperm(col, ldg)] = vv;
      int64_t row = row0 + crow + i;
      if (row < B) gout[row * ldg + col] = vv;
    }
}

__global__ __launch_bounds__(256) void fwd_train_kernel(FwdTrainArgs p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * ROWS;
  const int D = (int)p.D, H = (int)p.H;
  const int Dp = (D + 15) & ~15;
  const int lda_o = Dp + 4, lda_h = H + 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* sobs = (float*)smem;
  float* sh1 = sobs + ROWS * lda_o;
  float* sh2 = sh1 + ROWS * lda_h;
  for (int ix = tid; ix < ROWS * Dp; ix += 256) {
    int r = ix / Dp, d = ix % Dp;
    float vv = 0.0f;
    if (d < D && row0 + r < p.B) vv = p.obs[(row0 + r) * D + d];
    sobs[r * lda_o + kperm(d, Dp)] = vv;
  }
  __syncthreads();
  for (int c0 = w * 64; c0 < H; c0 += 256)
    chunk_fwd_train(sobs, lda_o, Dp, p.wt1, H, c0, p.b1, sh1, lda_h,
                    p.h1, row0, p.B, H, lane);
  __syncthreads();
  for (int c0 = w * 64; c0 < H; c0 += 256)
    chunk_fwd_train(sh1, lda_h, H, p.wt2, H, c0, p.b2, sh2, lda_h,
                    p.h2, row0, p.B, H, lane);
  __syncthreads();
  if (w == 0) {
    f32x4 acc[2] = {};
    const int kk = lane >> 4, bc = lane & 15;
    const float* arow = sh2 + (lane & 15) * lda_h + kk * (H >> 2);
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
        int64_t row = row0 + crow + i;
        if (row < p.B) p.head[row * HEADC + col] = acc[n][i] + p.bh[col];
      }
  }
}

// -------------------------------------------- masked PPO loss on head rows
// Like ppo_loss_kernel (nn_kernels.hip) but reads the raw head [B,HEADC]
// (logits cols [0,A), value col A), applies the action mask itself
// (invalid actions excluded from the softmax), and writes the fused
// gradient dhead [B,HEADC] (zeros for masked/pad columns).
struct PpoHeadArgs {
  int64_t B, A;
  float clip, vf_coef, ent_coef, inv_n, kl_coef;
  // Entropy floor (round 2): rows whose policy entropy has
  // collapsed below ent_floor get an EXTRA entropy bonus of
  // ent_floor_coef on top of ent_coef. A per-row hinge — exploration is
  // only forced where it has died, instead of globally inflating
  // ent_coef (which round-1 sweeps showed destabilizes training).
  float ent_floor, ent_floor_coef;
  const float* head;       // [B,HEADC]
  const uint8_t* mask;     // [B,A]
  const int64_t* actions;  // [B]
  const float* old_logp;
  const float* old_value;
  const float* adv;
  const float* ret;
  float* dhead;            // [B,HEADC] out
  float* stats;            // [4] sums
};

__global__ __launch_bounds__(256) void ppo_head_kernel(PpoHeadArgs a) {
  const int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int A = (int)a.A;
  float s_pi = 0.f, s_v = 0.f, s_ent = 0.f, s_cf = 0.f;
  for (int64_t i = i0; i < a.B; i += (int64_t)gridDim.x * blockDim.x) {
    // NOTE: do NOT stage the row into a local array — runtime-indexed
    // per-thread arrays go to scratch (5x); the row is L2/L1-resident
    // across the four passes anyway
    const float* lg = a.head + i * HEADC;
    const uint8_t* mk = a.mask + i * A;
    float* dlg = a.dhead + i * HEADC;
    const int act = (int)a.actions[i];
    float mx = -3.0e38f;
    for (int j = 0; j < A; ++j)
      if (mk[j]) mx = fmaxf(mx, lg[j]);
    float sum = 0.f;
    for (int j = 0; j < A; ++j)
      if (mk[j]) sum += __expf(lg[j] - mx);
    const float lsum = __logf(sum);
    const float logp_a = (lg[act] - mx) - lsum;
    float H = 0.f;
    for (int j = 0; j < A; ++j)
      if (mk[j]) {
        float lpj = (lg[j] - mx) - lsum;
        H -= __expf(lpj) * lpj;
      }
    const float advi = a.adv[i];
    const float ratio = __expf(logp_a - a.old_logp[i]);
    const float rcl = fminf(fmaxf(ratio, 1.f - a.clip), 1.f + a.clip);
    const float s1 = ratio * advi, s2 = rcl * advi;
    const float active = (s1 <= s2) ? 1.f : 0.f;
    // k3 KL-to-behavior penalty (ratio-1-log ratio >= 0); kl_coef=0 is
    // vanilla PPO. Folded into the pi stat; grad wrt logp is ratio-1.
    const float lr3 = logp_a - a.old_logp[i];
    s_pi += -fminf(s1, s2) + a.kl_coef * (ratio - 1.f - lr3);
    s_ent += H;
    s_cf += (fabsf(ratio - 1.f) > a.clip) ? 1.f : 0.f;
    const float g_logp = (-advi * ratio * active
                          + a.kl_coef * (ratio - 1.f)) * a.inv_n;
    const float ecoef = a.ent_coef
        + ((H < a.ent_floor) ? a.ent_floor_coef : 0.f);
    for (int j = 0; j < A; ++j) {
      float g = 0.f;
      if (mk[j]) {
        float lpj = (lg[j] - mx) - lsum;
        float pj = __expf(lpj);
        g = g_logp * (((j == act) ? 1.f : 0.f) - pj)
          + ecoef * pj * (lpj + H) * a.inv_n;
      }
      dlg[j] = g;
    }
    for (int j = A + 1; j < HEADC; ++j) dlg[j] = 0.f;
    const float v = lg[A], ov = a.old_value[i], rt = a.ret[i];
    const float vcl = ov + fminf(fmaxf(v - ov, -a.clip), a.clip);
    const float e1 = v - rt, e2 = vcl - rt;
    const float q1 = e1 * e1, q2 = e2 * e2;
    s_v += 0.5f * fmaxf(q1, q2);
    float dv;
    if (q1 >= q2) dv = e1;
    else dv = (fabsf(v - ov) < a.clip) ? e2 : 0.f;
    dlg[A] = a.vf_coef * dv * a.inv_n;
  }
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

// ------------------------------------- backward dX = (dY @ W^T) * tanh'(h)
// dY [B,K] (K = HEADC or H), W^T packed [K,N], h [B,N] saved activation.
struct BwdDxArgs {
  int64_t B, K, N;
  const float* dY;    // [B,K]
  const float* wtp;   // packed [K,N]
  const float* h;     // [B,N]
  float* dX;          // [B,N] out
};

__global__ __launch_bounds__(256) void bwd_dx_kernel(BwdDxArgs p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * ROWS;
  const int K = (int)p.K, N = (int)p.N;
  const int lda = K + 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* sdy = (float*)smem;    // [ROWS][K+4] kperm
  for (int ix = tid; ix < ROWS * K; ix += 256) {
    int r = ix / K, k = ix % K;
    float vv = 0.0f;
    if (row0 + r < p.B) vv = p.dY[(row0 + r) * K + k];
    sdy[r * lda + kperm(k, K)] = vv;
  }
  __syncthreads();
  for (int c0 = w * 64; c0 < N; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_core_16x64(sdy, lda, K, p.wtp, N, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        int64_t row = row0 + crow + i;
        if (row < p.B) {
          float g = acc[n][i];
          if (p.h) {           // fused tanh' (null h -> plain GEMM)
            float hv = p.h[row * N + col];
            g *= (1.0f - hv * hv);
          }
          p.dX[row * N + col] = g;
        }
      }
  }
}

// ----------------------------------------- backward dW += X^T @ dY over B
// X [B,M], dY [B,N] -> dW [M,N] (f32 atomicAdd; caller zeroes dW).
// Block: 64(m) x 64(n) C-tile x BC-deep batch chunk; both operands staged
// transposed into kperm LDS tiles ([64 c-dim][64 kperm] padded).
struct BwdDwArgs {
  int64_t B, M, N, BC;   // BC: batch rows per block (multiple of 64)
  const float* X;        // [B,M]
  const float* dY;       // [B,N]
  float* dW;             // [M,N] accumulated
};

#define DW_KC 128      // k-depth per staged chunk
#define DW_LD (DW_KC + 4)  // row stride: % 64 == 4 -> distinct bank groups

__global__ __launch_bounds__(256) void bwd_dw_kernel(BwdDwArgs p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int M = (int)p.M, N = (int)p.N;
  const int mtiles = (M + 63) >> 6, ntiles = (N + 63) >> 6;
  int bid = blockIdx.x;
  const int m0 = (bid % mtiles) * 64; bid /= mtiles;
  const int n0 = (bid % ntiles) * 64; bid /= ntiles;
  const int64_t b0 = (int64_t)bid * p.BC;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xt = (float*)smem;          // [64 m][DW_LD] kperm over k (128-deep)
  float* yt = xt + 64 * DW_LD;       // [64 n][DW_LD]
  f32x4 acc[4] = {};
  const int wm = w * 16;             // wave's 16 m-rows within the tile
  const int ar = lane & 15, kk = lane >> 4, bc = lane & 15;
  for (int kc = 0; kc < (int)p.BC; kc += DW_KC) {
    // stage X^T and dY^T chunks with float4 global loads: thread ix covers
    // (k, j4): 4 consecutive features j of batch row b -> 4 LDS rows at
    // column kperm(k). 128 k x 16 j-quads = 2048 quads / 256 threads.
    for (int ix = tid; ix < DW_KC * 16; ix += 256) {
      int k = ix >> 4, j4 = (ix & 15) * 4;
      int64_t b = b0 + kc + k;
      int kp = kperm(k, DW_KC);
      f32x4 xv = {}, yv = {};
      if (b < p.B) {
        if (m0 + j4 + 3 < M) xv = *(const f32x4*)(p.X + b * M + m0 + j4);
        else
          for (int t = 0; t < 4; ++t)
            if (m0 + j4 + t < M) xv[t] = p.X[b * M + m0 + j4 + t];
        if (n0 + j4 + 3 < N) yv = *(const f32x4*)(p.dY + b * N + n0 + j4);
        else
          for (int t = 0; t < 4; ++t)
            if (n0 + j4 + t < N) yv[t] = p.dY[b * N + n0 + j4 + t];
      }
      #pragma unroll
      for (int t = 0; t < 4; ++t) {
        xt[(j4 + t) * DW_LD + kp] = xv[t];
        yt[(j4 + t) * DW_LD + kp] = yv[t];
      }
    }
    __syncthreads();
    // C[wm..wm+15][0..63] over the 128-deep chunk; fragments are float4
    // LDS reads (kperm: k = 16*k16 + 4s + kk -> kk*(KC/4) + 4*k16 + s)
    const float* arow = xt + (wm + ar) * DW_LD + kk * (DW_KC / 4);
    const float* brows = yt + bc * DW_LD + kk * (DW_KC / 4);
    #pragma unroll
    for (int k16 = 0; k16 < DW_KC / 16; ++k16) {
      f32x4 av = *(const f32x4*)(arow + 4 * k16);
      #pragma unroll
      for (int n = 0; n < 4; ++n) {
        f32x4 bv = *(const f32x4*)(brows + n * 16 * DW_LD + 4 * k16);
        #pragma unroll
        for (int s = 0; s < 4; ++s)
          acc[n] = __builtin_amdgcn_mfma_f32_16x16x4f32(av[s], bv[s], acc[n],
                                                        0, 0, 0);
      }
    }
    __syncthreads();
  }
  const int crow = (lane >> 4) * 4, ccol = lane & 15;
  #pragma unroll
  for (int n = 0; n < 4; ++n)
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int m = m0 + wm + crow + i;
      int nn = n0 + n * 16 + ccol;
      if (m < M && nn < N) atomicAdd(&p.dW[(int64_t)m * N + nn], acc[n][i]);
    }
}

// ------------------------------------------------------- bias column sums
// db[N] += sum_b dY[b][n] (caller zeroes db). Two-level: each block owns a
// 256-column tile x one batch chunk (coalesced row reads), one atomicAdd
// per column per chunk.
struct ColSumArgs {
  int64_t B, N, BCS;   // batch rows per chunk
  const float* dY;
  float* db;
};

__global__ __launch_bounds__(256) void colsum_kernel(ColSumArgs p) {
  int64_t n = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= p.N) return;
  int64_t b0 = (int64_t)blockIdx.y * p.BCS;
  int64_t b1 = b0 + p.BCS < p.B ? b0 + p.BCS : p.B;
  float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
  int64_t b = b0;
  for (; b + 3 < b1; b += 4) {   // 4 independent loads in flight
    s0 += p.dY[b * p.N + n];
    s1 += p.dY[(b + 1) * p.N + n];
    s2 += p.dY[(b + 2) * p.N + n];
    s3 += p.dY[(b + 3) * p.N + n];
  }
  for (; b < b1; ++b) s0 += p.dY[b * p.N + n];
  atomicAdd(&p.db[n], (s0 + s1) + (s2 + s3));
}

}  // namespace

extern "C" int rlsched_fwd_train(const int64_t* ip, const uint64_t* bufs,
                                 void* stream) {
  FwdTrainArgs p;
  p.B = ip[0]; p.D = ip[1]; p.H = ip[2];
  if (p.H % 64) return -1;
  p.obs = (const float*)bufs[0];
  p.wt1 = (const float*)bufs[1];
  p.b1 = (const float*)bufs[2];
  p.wt2 = (const float*)bufs[3];
  p.b2 = (const float*)bufs[4];
  p.wth = (const float*)bufs[5];
  p.bh = (const float*)bufs[6];
  p.h1 = (float*)bufs[7];
  p.h2 = (float*)bufs[8];
  p.head = (float*)bufs[9];
  const int Dp = ((int)p.D + 15) & ~15;
  size_t lds = sizeof(float) * (ROWS * (Dp + 4) + 2 * ROWS * ((int)p.H + 4));
  dim3 grid((unsigned)((p.B + ROWS - 1) / ROWS));
  hipLaunchKernelGGL(fwd_train_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

extern "C" int rlsched_ppo_head(const int64_t* ip, const double* fp,
                                const uint64_t* bufs, void* stream) {
  PpoHeadArgs a;
  a.B = ip[0]; a.A = ip[1];
  a.clip = (float)fp[0]; a.vf_coef = (float)fp[1];
  a.ent_coef = (float)fp[2]; a.inv_n = (float)fp[3];
  a.kl_coef = (float)fp[4];
  a.ent_floor = (float)fp[5]; a.ent_floor_coef = (float)fp[6];
  a.head = (const float*)bufs[0];
  a.mask = (const uint8_t*)bufs[1];
  a.actions = (const int64_t*)bufs[2];
  a.old_logp = (const float*)bufs[3];
  a.old_value = (const float*)bufs[4];
  a.adv = (const float*)bufs[5];
  a.ret = (const float*)bufs[6];
  a.dhead = (float*)bufs[7];
  a.stats = (float*)bufs[8];
  int grid = (int)((a.B + 255) / 256);
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(ppo_head_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, a);
  return (int)hipGetLastError();
}

extern "C" int rlsched_bwd_dx(const int64_t* ip, const uint64_t* bufs,
                              void* stream) {
  BwdDxArgs p;
  p.B = ip[0]; p.K = ip[1]; p.N = ip[2];
  if (p.K % 16 || p.N % 64) return -1;
  p.dY = (const float*)bufs[0];
  p.wtp = (const float*)bufs[1];
  p.h = (const float*)bufs[2];
  p.dX = (float*)bufs[3];
  size_t lds = sizeof(float) * ROWS * ((int)p.K + 4);
  dim3 grid((unsigned)((p.B + ROWS - 1) / ROWS));
  hipLaunchKernelGGL(bwd_dx_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

extern "C" int rlsched_bwd_dw(const int64_t* ip, const uint64_t* bufs,
                              void* stream) {
  BwdDwArgs p;
  p.B = ip[0]; p.M = ip[1]; p.N = ip[2]; p.BC = ip[3];
  if (p.BC % DW_KC) return -1;  // batch-row guards handle the tail
  p.X = (const float*)bufs[0];
  p.dY = (const float*)bufs[1];
  p.dW = (float*)bufs[2];
  const int mtiles = ((int)p.M + 63) >> 6, ntiles = ((int)p.N + 63) >> 6;
  const int chunks = (int)((p.B + p.BC - 1) / p.BC);
  size_t lds = sizeof(float) * 2 * 64 * DW_LD;
  dim3 grid((unsigned)(mtiles * ntiles * chunks));
  hipLaunchKernelGGL(bwd_dw_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

extern "C" int rlsched_colsum(const int64_t* ip, const uint64_t* bufs,
                              void* stream) {
  ColSumArgs p;
  p.B = ip[0]; p.N = ip[1];
  p.BCS = 512;
  p.dY = (const float*)bufs[0];
  p.db = (float*)bufs[1];
  unsigned chunks = (unsigned)((p.B + p.BCS - 1) / p.BCS);
  dim3 grid((unsigned)((p.N + 255) / 256), chunks);
  hipLaunchKernelGGL(colsum_kernel, grid, dim3(256), 0, (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

// ---------------------------------------------- fused minibatch gather
// One launch gathers all seven per-sample arrays for a PPO minibatch
// (replaces 7 torch index_select kernels). One thread per row; float4
// obs copies (D % 4 == 0).
namespace {

struct GatherArgs {
  int64_t MB, D, A;
  const int64_t* ix;       // [MB]
  const float* obs;        // [Bsrc,D]
  const uint8_t* mask;     // [Bsrc,A]
  const int64_t* acts;     // [Bsrc]
  const float* olp;
  const float* ov;
  const float* adv;
  const float* ret;
  float* o_obs;            // [MB,D]
  uint8_t* o_mask;         // [MB,A]
  int64_t* o_acts;
  float* o_olp;
  float* o_ov;
  float* o_adv;
  float* o_ret;
};

__global__ __launch_bounds__(256) void gather_kernel(GatherArgs g) {
  // 32 lanes cooperate per row: the row's float4s load coalesced instead
  // of one thread walking 21 cachelines serially
  const int sub = threadIdx.x & 31;
  int64_t r = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 5;
  if (r >= g.MB) return;
  int64_t s = g.ix[r];
  const int D = (int)g.D, A = (int)g.A;
  const f32x4* src = (const f32x4*)(g.obs + s * D);
  f32x4* dst = (f32x4*)(g.o_obs + r * D);
  for (int d = sub; d < D / 4; d += 32) dst[d] = src[d];
  for (int a = sub; a < A; a += 32) g.o_mask[r * A + a] = g.mask[s * A + a];
  if (sub == 0) {
    g.o_acts[r] = g.acts[s];
    g.o_olp[r] = g.olp[s];
    g.o_ov[r] = g.ov[s];
    g.o_adv[r] = g.adv[s];
    g.o_ret[r] = g.ret[s];
  }
}

}  // namespace

extern "C" int rlsched_gather_mb(const int64_t* ip, const uint64_t* bufs,
                                 void* stream) {
  GatherArgs g;
  g.MB = ip[0]; g.D = ip[1]; g.A = ip[2];
  if (g.D % 4) return -1;
  g.ix = (const int64_t*)bufs[0];
  g.obs = (const float*)bufs[1];
  g.mask = (const uint8_t*)bufs[2];
  g.acts = (const int64_t*)bufs[3];
  g.olp = (const float*)bufs[4];
  g.ov = (const float*)bufs[5];
  g.adv = (const float*)bufs[6];
  g.ret = (const float*)bufs[7];
  g.o_obs = (float*)bufs[8];
  g.o_mask = (uint8_t*)bufs[9];
  g.o_acts = (int64_t*)bufs[10];
  g.o_olp = (float*)bufs[11];
  g.o_ov = (float*)bufs[12];
  g.o_adv = (float*)bufs[13];
  g.o_ret = (float*)bufs[14];
  dim3 grid((unsigned)((g.MB * 32 + 255) / 256));
  hipLaunchKernelGGL(gather_kernel, grid, dim3(256), 0, (hipStream_t)stream, g);
  return (int)hipGetLastError();
}

// ------------------------------------------------ fused clip + Adam step
// Consumes the FLAT grad buffer in OUR layout (dW [in,out], biases) and
// updates the torch parameters in THEIR layout ([out,in]) directly —
// replaces per-minibatch grad-assembly copies + torch clip + foreach-Adam
// (~15 launches) with adam_bump (bias corrections, 1 thread) + adam_step.
// m/v state is indexed in flat-grad order. Grad clip: the caller computes
// coef = min(1, max_norm/(||g||+1e-6)) into a device scalar (torch norm).
// Numerics match torch.optim.Adam: denom = sqrt(v)/sqrt(bc2) + eps,
// p -= (lr/bc1) * m / denom.
namespace {

#define ADAM_MAX_SEG 12

struct AdamArgs {
  int64_t nseg;
  float lr, beta1, beta2, eps;
  const float* gflat;
  float* m;
  float* v;
  const float* coef;     // [1] clip coefficient
  const float* bc;       // [2] bias corrections (from adam_bump)
  // per segment: grads at gflat[goff + r*C + c] for r<R, c0<=c<c1;
  // m/v at m[moff + (r*(c1-c0) + (c-c0))];
  // param element: transposed ? param[(c-c0)*R + r] : param[r*C + c] --
  // for column-sliced heads the param row index is (c-c0).
  int64_t goff[ADAM_MAX_SEG], moff[ADAM_MAX_SEG];
  int64_t R[ADAM_MAX_SEG], C[ADAM_MAX_SEG];
  int64_t c0[ADAM_MAX_SEG], c1[ADAM_MAX_SEG];
  int64_t cum[ADAM_MAX_SEG + 1];   // cumulative element counts
  int64_t rmsprop;                 // 0 = Adam, 1 = RMSprop (alpha=beta2)
  float* param[ADAM_MAX_SEG];
  int64_t transposed[ADAM_MAX_SEG];
};

struct AdamState {
  float* t;     // [1] step count (fp32; exact for < 2^24 steps)
  float* bc;    // [2] out: 1-b1^t, 1-b2^t
  float beta1, beta2;
};

__global__ void adam_bump_kernel(AdamState s) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    float t = s.t[0] + 1.0f;
    s.t[0] = t;
    s.bc[0] = 1.0f - __powf(s.beta1, t);
    s.bc[1] = 1.0f - __powf(s.beta2, t);
  }
}

__global__ __launch_bounds__(256) void adam_step_kernel(AdamArgs a) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= a.cum[a.nseg]) return;
  int seg = 0;
  while (i >= a.cum[seg + 1]) ++seg;
  int64_t e = i - a.cum[seg];
  const int64_t W = a.c1[seg] - a.c0[seg];
  const int64_t r = e / W, cc = e % W;
  const int64_t c = a.c0[seg] + cc;
  float g = a.gflat[a.goff[seg] + r * a.C[seg] + c] * a.coef[0];
  const int64_t mi = a.moff[seg] + e;
  float upd;
  if (a.rmsprop) {   // torch.optim.RMSprop (momentum 0, uncentered)
    float v = a.beta2 * a.v[mi] + (1.0f - a.beta2) * g * g;
    a.v[mi] = v;
    upd = a.lr * g / (__fsqrt_rn(v) + a.eps);
  } else {
    float m = a.beta1 * a.m[mi] + (1.0f - a.beta1) * g;
    float v = a.beta2 * a.v[mi] + (1.0f - a.beta2) * g * g;
    a.m[mi] = m;
    a.v[mi] = v;
    const float denom = __fsqrt_rn(v) / __fsqrt_rn(a.bc[1]) + a.eps;
    upd = (a.lr / a.bc[0]) * m / denom;
  }
  float* p = a.param[seg];
  // param layouts: transposed weights [outW, R]; sliced/non-transposed
  // segments are compact [R, W] (W = c1-c0; biases have R == 1)
  if (a.transposed[seg]) p[cc * a.R[seg] + r] -= upd;
  else p[r * W + cc] -= upd;
}

}  // namespace

extern "C" int rlsched_adam_bump(const int64_t* ip, const double* fp,
                                 const uint64_t* bufs, void* stream) {
  AdamState s;
  (void)ip;
  s.beta1 = (float)fp[0];
  s.beta2 = (float)fp[1];
  s.t = (float*)bufs[0];
  s.bc = (float*)bufs[1];
  hipLaunchKernelGGL(adam_bump_kernel, dim3(1), dim3(1), 0,
                     (hipStream_t)stream, s);
  return (int)hipGetLastError();
}

extern "C" int rlsched_adam_step(const int64_t* ip, const double* fp,
                                 const uint64_t* bufs, void* stream) {
  AdamArgs a;
  a.nseg = ip[0];
  if (a.nseg > ADAM_MAX_SEG) return -1;
  a.lr = (float)fp[0]; a.beta1 = (float)fp[1];
  a.beta2 = (float)fp[2]; a.eps = (float)fp[3];
  a.gflat = (const float*)bufs[0];
  a.m = (float*)bufs[1];
  a.v = (float*)bufs[2];
  a.coef = (const float*)bufs[3];
  a.bc = (const float*)bufs[4];
  // per-seg metadata follows: ip[1 + 6*s .. ], param ptrs at bufs[5 + s]
  a.cum[0] = 0;
  for (int s = 0; s < a.nseg; ++s) {
    a.goff[s] = ip[1 + 6 * s + 0];
    a.moff[s] = ip[1 + 6 * s + 1];
    a.R[s] = ip[1 + 6 * s + 2];
    a.C[s] = ip[1 + 6 * s + 3];
    a.c0[s] = ip[1 + 6 * s + 4];
    a.c1[s] = ip[1 + 6 * s + 5];
    a.transposed[s] = ip[1 + 6 * a.nseg + s];
    a.param[s] = (float*)bufs[5 + s];
    a.cum[s + 1] = a.cum[s] + a.R[s] * (a.c1[s] - a.c0[s]);
  }
  a.rmsprop = ip[1 + 7 * a.nseg];
  dim3 grid((unsigned)((a.cum[a.nseg] + 255) / 256));
  hipLaunchKernelGGL(adam_step_kernel, grid, dim3(256), 0,
                     (hipStream_t)stream, a);
  return (int)hipGetLastError();
}

// --------------------------- bf16 mixed-precision update-path kernels
// Opt-in (bench --update-dtype bf16): forward and backward GEMMs on
// mfma_f32_16x16x32_bf16 (~13x the f32 rate), f32 accumulation, f32 loss
// head and f32 weight gradients/Adam (activations h1/h2 stored bf16 —
// half the save/reload traffic). Numerics: mixed-precision class; the
// fp32 path stays the exactness default.
namespace {

struct FwdTrainBf16Args {
  int64_t B, D, H;
  const float* obs;
  const rls_bf16* wt1;   // packed8 [Dp,H]
  const float* b1;
  const rls_bf16* wt2;   // packed8 [H,H]
  const float* b2;
  const rls_bf16* wth;   // packed8 [H,HEADC]
  const float* bh;
  rls_bf16* h1;          // [B,H] bf16 out
  rls_bf16* h2;          // [B,H] bf16 out
  float* head;           // [B,HEADC] f32 out
};

__global__ __launch_bounds__(256) void fwd_train_bf16_kernel(
    FwdTrainBf16Args p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * ROWS;
  const int D = (int)p.D, H = (int)p.H;
  const int Dp = (D + 31) & ~31;
  const int lda_o = Dp + 8, lda_h = H + 8;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  rls_bf16* sobs = (rls_bf16*)smem;
  rls_bf16* sh1 = sobs + ROWS * lda_o;
  rls_bf16* sh2 = sh1 + ROWS * lda_h;
  {
    typedef rls_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    for (int ix = tid; ix < ROWS * (Dp / 4); ix += 256) {
      int r = ix / (Dp / 4), d4 = (ix % (Dp / 4)) * 4;
      bf16x4 bv = {};
      if (row0 + r < p.B) {
        if (d4 + 3 < D) {
          f32x4 v = *(const f32x4*)(p.obs + (row0 + r) * D + d4);
          #pragma unroll
          for (int t = 0; t < 4; ++t) bv[t] = (rls_bf16)v[t];
        } else {
          for (int t = 0; t < 4; ++t)
            if (d4 + t < D)
              bv[t] = (rls_bf16)p.obs[(row0 + r) * D + d4 + t];
        }
      }
      *(bf16x4*)(sobs + r * lda_o + d4) = bv;
    }
  }
  __syncthreads();
  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_bf16_core_16x64(sobs, lda_o, Dp, p.wt1, H, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        rls_bf16 vv = (rls_bf16)tanhf(acc[n][i] + p.b1[col]);
        sh1[(crow + i) * lda_h + col] = vv;
        if (row0 + crow + i < p.B) p.h1[(row0 + crow + i) * H + col] = vv;
      }
  }
  __syncthreads();
  for (int c0 = w * 64; c0 < H; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_bf16_core_16x64(sh1, lda_h, H, p.wt2, H, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        rls_bf16 vv = (rls_bf16)tanhf(acc[n][i] + p.b2[col]);
        sh2[(crow + i) * lda_h + col] = vv;
        if (row0 + crow + i < p.B) p.h2[(row0 + crow + i) * H + col] = vv;
      }
  }
  __syncthreads();
  if (w == 0) {
    f32x4 acc[2] = {};
    const int ar = lane & 15, kk = lane >> 4, bc = lane & 15;
    const rls_bf16* arow = sh2 + ar * lda_h + kk * 8;
    const rls_bf16* bbase = p.wth + (int64_t)8 * (kk * HEADC + bc);
    for (int k32 = 0; k32 < (H >> 5); ++k32) {
      bf16x8 av = *(const bf16x8*)(arow + 32 * k32);
      const rls_bf16* bk = bbase + (int64_t)32 * k32 * HEADC;
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
        if (row0 + crow + i < p.B)
          p.head[(row0 + crow + i) * HEADC + col] = acc[n][i] + p.bh[col];
      }
  }
}

// dX = (dY @ W^T) [* (1 - h^2)]: dY f32 staged to bf16, W packed8 bf16,
// h bf16 (null -> plain GEMM), dX f32.
struct BwdDxBf16Args {
  int64_t B, K, N;
  const float* dY;
  const rls_bf16* wtp;
  const rls_bf16* h;
  float* dX;
};

__global__ __launch_bounds__(256) void bwd_dx_bf16_kernel(BwdDxBf16Args p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * ROWS;
  const int K = (int)p.K, N = (int)p.N;
  const int lda = K + 8;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  rls_bf16* sdy = (rls_bf16*)smem;    // [ROWS][K+8] plain
  // vectorized staging: one float4 dY load + one 8-byte bf16x4 store per
  // (row, 4-col) slot (K % 4 == 0; contiguous-k LDS layout)
  typedef rls_bf16 bf16x4 __attribute__((ext_vector_type(4)));
  for (int ix = tid; ix < ROWS * (K / 4); ix += 256) {
    int r = ix / (K / 4), k4 = (ix % (K / 4)) * 4;
    f32x4 vv = {};
    if (row0 + r < p.B) vv = *(const f32x4*)(p.dY + (row0 + r) * K + k4);
    bf16x4 bv;
    #pragma unroll
    for (int t = 0; t < 4; ++t) bv[t] = (rls_bf16)vv[t];
    *(bf16x4*)(sdy + r * lda + k4) = bv;
  }
  __syncthreads();
  for (int c0 = w * 64; c0 < N; c0 += 256) {
    f32x4 acc[4] = {};
    mfma_bf16_core_16x64(sdy, lda, K, p.wtp, N, c0, lane, acc);
    const int crow = (lane >> 4) * 4, ccol = lane & 15;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int col = c0 + n * 16 + ccol;
        int64_t row = row0 + crow + i;
        if (row < p.B) {
          float g = acc[n][i];
          if (p.h) {
            float hv = (float)p.h[row * N + col];
            g *= (1.0f - hv * hv);
          }
          p.dX[row * N + col] = g;
        }
      }
  }
}

// dW[M,N] += X^T @ dY over batch: operands staged TRANSPOSED into bf16
// LDS tiles ([64 features][DW_KC k] plain contiguous-k layout), bf16 MFMA,
// f32 atomic accumulation. X may be bf16 (saved activations) or f32 (obs);
// dY f32 (upstream grads), converted during staging.
struct BwdDwBf16Args {
  int64_t B, M, N, BC;
  const float* Xf;          // one of Xf/Xb is null
  const rls_bf16* Xb;
  const float* dY;
  float* dW;
};

#define DWB_KC 256           // bf16 k-chunk (2x f32: LDS is half the bytes)
#define DWB_LD (DWB_KC + 8)  // bf16 elements; /2 dwords % 64 == 4

__global__ __launch_bounds__(256) void bwd_dw_bf16_kernel(BwdDwBf16Args p) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int w = tid >> 6;
  const int M = (int)p.M, N = (int)p.N;
  const int mtiles = (M + 63) >> 6, ntiles = (N + 63) >> 6;
  int bid = blockIdx.x;
  const int m0 = (bid % mtiles) * 64; bid /= mtiles;
  const int n0 = (bid % ntiles) * 64; bid /= ntiles;
  const int64_t b0 = (int64_t)bid * p.BC;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  rls_bf16* xt = (rls_bf16*)smem;          // [64 m][DWB_LD]
  rls_bf16* yt = xt + 64 * DWB_LD;         // [64 n][DWB_LD]
  f32x4 acc[4] = {};
  const int wm = w * 16;
  const int ar = lane & 15, kk = lane >> 4, bc = lane & 15;
  for (int kc = 0; kc < (int)p.BC; kc += DWB_KC) {
    typedef rls_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    for (int ix = tid; ix < DWB_KC * 16; ix += 256) {
      int k = ix >> 4, j4 = (ix & 15) * 4;
      int64_t b = b0 + kc + k;
      rls_bf16 xv[4] = {}, yv[4] = {};
      if (b < p.B) {
        if (m0 + j4 + 3 < M) {
          if (p.Xf) {
            f32x4 v = *(const f32x4*)(p.Xf + b * M + m0 + j4);
            #pragma unroll
            for (int t = 0; t < 4; ++t) xv[t] = (rls_bf16)v[t];
          } else {
            *(bf16x4*)xv = *(const bf16x4*)(p.Xb + b * M + m0 + j4);
          }
        } else {
          for (int t = 0; t < 4; ++t)
            if (m0 + j4 + t < M)
              xv[t] = (rls_bf16)(p.Xf ? p.Xf[b * M + m0 + j4 + t]
                                      : (float)p.Xb[b * M + m0 + j4 + t]);
        }
        if (n0 + j4 + 3 < N) {
          f32x4 v = *(const f32x4*)(p.dY + b * N + n0 + j4);
          #pragma unroll
          for (int t = 0; t < 4; ++t) yv[t] = (rls_bf16)v[t];
        } else {
          for (int t = 0; t < 4; ++t)
            if (n0 + j4 + t < N) yv[t] = (rls_bf16)p.dY[b * N + n0 + j4 + t];
        }
      }
      #pragma unroll
      for (int t = 0; t < 4; ++t) {
        xt[(j4 + t) * DWB_LD + k] = xv[t];
        yt[(j4 + t) * DWB_LD + k] = yv[t];
      }
    }
    __syncthreads();
    const rls_bf16* arow = xt + (wm + ar) * DWB_LD + kk * 8;
    const rls_bf16* brows = yt + bc * DWB_LD + kk * 8;
    #pragma unroll
    for (int k32 = 0; k32 < DWB_KC / 32; ++k32) {
      bf16x8 av = *(const bf16x8*)(arow + 32 * k32);
      #pragma unroll
      for (int n = 0; n < 4; ++n) {
        bf16x8 bv = *(const bf16x8*)(brows + n * 16 * DWB_LD + 32 * k32);
        acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc[n],
                                                         0, 0, 0);
      }
    }
    __syncthreads();
  }
  const int crow = (lane >> 4) * 4, ccol = lane & 15;
  #pragma unroll
  for (int n = 0; n < 4; ++n)
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int m = m0 + wm + crow + i;
      int nn = n0 + n * 16 + ccol;
      if (m < M && nn < N) atomicAdd(&p.dW[(int64_t)m * N + nn], acc[n][i]);
    }
}

}  // namespace

extern "C" int rlsched_fwd_train_bf16(const int64_t* ip, const uint64_t* bufs,
                                      void* stream) {
  FwdTrainBf16Args p;
  p.B = ip[0]; p.D = ip[1]; p.H = ip[2];
  if (p.H % 64) return -1;
  p.obs = (const float*)bufs[0];
  p.wt1 = (const rls_bf16*)bufs[1];
  p.b1 = (const float*)bufs[2];
  p.wt2 = (const rls_bf16*)bufs[3];
  p.b2 = (const float*)bufs[4];
  p.wth = (const rls_bf16*)bufs[5];
  p.bh = (const float*)bufs[6];
  p.h1 = (rls_bf16*)bufs[7];
  p.h2 = (rls_bf16*)bufs[8];
  p.head = (float*)bufs[9];
  const int Dp = ((int)p.D + 31) & ~31;
  size_t lds = sizeof(rls_bf16)
             * (ROWS * (Dp + 8) + 2 * ROWS * ((int)p.H + 8));
  dim3 grid((unsigned)((p.B + ROWS - 1) / ROWS));
  hipLaunchKernelGGL(fwd_train_bf16_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

extern "C" int rlsched_bwd_dx_bf16(const int64_t* ip, const uint64_t* bufs,
                                   void* stream) {
  BwdDxBf16Args p;
  p.B = ip[0]; p.K = ip[1]; p.N = ip[2];
  if (p.K % 32 || p.N % 64) return -1;
  p.dY = (const float*)bufs[0];
  p.wtp = (const rls_bf16*)bufs[1];
  p.h = (const rls_bf16*)bufs[2];
  p.dX = (float*)bufs[3];
  size_t lds = sizeof(rls_bf16) * ROWS * ((int)p.K + 8);
  dim3 grid((unsigned)((p.B + ROWS - 1) / ROWS));
  hipLaunchKernelGGL(bwd_dx_bf16_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

extern "C" int rlsched_bwd_dw_bf16(const int64_t* ip, const uint64_t* bufs,
                                   void* stream) {
  BwdDwBf16Args p;
  p.B = ip[0]; p.M = ip[1]; p.N = ip[2]; p.BC = ip[3];
  const int64_t x_is_bf16 = ip[4];
  if (p.BC % DWB_KC) return -1;
  p.Xf = x_is_bf16 ? nullptr : (const float*)bufs[0];
  p.Xb = x_is_bf16 ? (const rls_bf16*)bufs[0] : nullptr;
  p.dY = (const float*)bufs[1];
  p.dW = (float*)bufs[2];
  const int mtiles = ((int)p.M + 63) >> 6, ntiles = ((int)p.N + 63) >> 6;
  const int chunks = (int)((p.B + p.BC - 1) / p.BC);
  size_t lds = sizeof(rls_bf16) * 2 * 64 * DWB_LD;
  dim3 grid((unsigned)(mtiles * ntiles * chunks));
  hipLaunchKernelGGL(bwd_dw_bf16_kernel, grid, dim3(256), lds,
                     (hipStream_t)stream, p);
  return (int)hipGetLastError();
}

// ------------------------------------------- IMPALA loss head (fwd+bwd)
// Per sample: masked softmax over head logits, logp(a), entropy, and the
// IMPALA objective  L = -pg_adv*logp + 0.5*vf*(v - vs)^2 - ent*H  with
// dhead written in one pass (pg_adv and vs come from the V-trace scan,
// which itself consumes logp computed HERE — so the learner calls this
// kernel twice: pass 1 (logp_only) fills lp_cur for the scan, pass 2
// computes the loss/grads with the scan's outputs).
namespace {

struct VtraceHeadArgs {
  int64_t B, A;
  int64_t logp_only;
  float vf_coef, ent_coef, inv_n;
  const float* head;       // [B,HEADC]
  const uint8_t* mask;     // [B,A]
  const int64_t* actions;  // [B]
  const float* pg_adv;     // [B] (pass 2)
  const float* vs;         // [B] (pass 2)
  float* lp_cur;           // [B] out (pass 1) / in unused
  float* value;            // [B] out (pass 1): V(x) under current params
  float* dhead;            // [B,HEADC] out (pass 2)
  float* stats;            // [4]: pi_loss, v_loss, entropy, 0
};

__global__ __launch_bounds__(256) void vtrace_head_kernel(VtraceHeadArgs a) {
  const int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int A = (int)a.A;
  float s_pi = 0.f, s_v = 0.f, s_ent = 0.f;
  for (int64_t i = i0; i < a.B; i += (int64_t)gridDim.x * blockDim.x) {
    const float* lg = a.head + i * HEADC;
    const uint8_t* mk = a.mask + i * A;
    const int act = (int)a.actions[i];
    float mx = -3.0e38f;
    for (int j = 0; j < A; ++j)
      if (mk[j]) mx = fmaxf(mx, lg[j]);
    float sum = 0.f;
    for (int j = 0; j < A; ++j)
      if (mk[j]) sum += __expf(lg[j] - mx);
    const float lsum = __logf(sum);
    const float logp_a = (lg[act] - mx) - lsum;
    if (a.logp_only) {
      a.lp_cur[i] = logp_a;
      a.value[i] = lg[A];
      continue;
    }
    float H = 0.f;
    for (int j = 0; j < A; ++j)
      if (mk[j]) {
        float lpj = (lg[j] - mx) - lsum;
        H -= __expf(lpj) * lpj;
      }
    const float adv = a.pg_adv[i];
    s_pi += -adv * logp_a;
    s_ent += H;
    const float g_logp = -adv * a.inv_n;
    float* dlg = a.dhead + i * HEADC;
    for (int j = 0; j < A; ++j) {
      float g = 0.f;
      if (mk[j]) {
        float lpj = (lg[j] - mx) - lsum;
        float pj = __expf(lpj);
        g = g_logp * (((j == act) ? 1.f : 0.f) - pj)
          + a.ent_coef * pj * (lpj + H) * a.inv_n;
      }
      dlg[j] = g;
    }
    for (int j = A + 1; j < HEADC; ++j) dlg[j] = 0.f;
    const float verr = lg[A] - a.vs[i];
    s_v += 0.5f * verr * verr;
    dlg[A] = a.vf_coef * verr * a.inv_n;
  }
  __shared__ float red[3][256 / WAVE];
  float vals[3] = {s_pi, s_v, s_ent};
  const int lane = threadIdx.x & (WAVE - 1), w = threadIdx.x >> 6;
  #pragma unroll
  for (int k = 0; k < 3; ++k) {
    float x = vals[k];
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
    if (lane == 0) red[k][w] = x;
  }
  __syncthreads();
  if (threadIdx.x < 3) {
    float x = 0.f;
    for (int q = 0; q < 256 / WAVE; ++q) x += red[threadIdx.x][q];
    atomicAdd(&a.stats[threadIdx.x], x);
  }
}

}  // namespace

extern "C" int rlsched_vtrace_head(const int64_t* ip, const double* fp,
                                   const uint64_t* bufs, void* stream) {
  VtraceHeadArgs a;
  a.B = ip[0]; a.A = ip[1]; a.logp_only = ip[2];
  a.vf_coef = (float)fp[0]; a.ent_coef = (float)fp[1];
  a.inv_n = (float)fp[2];
  a.head = (const float*)bufs[0];
  a.mask = (const uint8_t*)bufs[1];
  a.actions = (const int64_t*)bufs[2];
  a.pg_adv = (const float*)bufs[3];
  a.vs = (const float*)bufs[4];
  a.lp_cur = (float*)bufs[5];
  a.value = (float*)bufs[6];
  a.dhead = (float*)bufs[7];
  a.stats = (float*)bufs[8];
  int grid = (int)((a.B + 255) / 256);
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(vtrace_head_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, a);
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
