// Standalone probe: recover the A/B fragment layout of
// __builtin_amdgcn_mfma_f32_16x16x32_bf16 on gfx950 by testing candidate
// lane->(row,k) maps against a host reference. Build & run on a GPU box:
//   hipcc --offload-arch=gfx950 -O2 tools/probe_bf16_mfma.cpp -o /tmp/probe && /tmp/probe
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cmath>

typedef __bf16 bf16;
typedef bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

// candidate k-maps for element j (0..7) of lane l (kk = l>>4):
__device__ int kmap(int cand, int kk, int j) {
  switch (cand) {
    case 0: return kk * 8 + j;              // contiguous 8
    case 1: return kk * 4 + (j & 3) + 16 * (j >> 2);  // two 16-halves
    case 2: return kk + 4 * j;              // strided 4
    case 3: return j * 4 + kk;              // same as 2? no: k= 4j+kk
    case 4: return kk * 2 + (j & 1) + 8 * (j >> 1);   // pairs per 8-block
    default: return 0;
  }
}

__global__ void probe(const float* A, const float* B, float* D, int cand) {
  int l = threadIdx.x;
  int row = l & 15, kk = l >> 4;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    int k = kmap(cand, kk, j);
    a[j] = (bf16)A[row * 32 + k];
    b[j] = (bf16)B[k * 16 + row];   // B[k][col]: col = l&15
  }
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  // C/D: col = l&15, row = (l>>4)*4 + i
  for (int i = 0; i < 4; ++i)
    D[(kk * 4 + i) * 16 + row] = acc[i];
}

int main() {
  float hA[16 * 32], hB[32 * 16], hD[256], ref[256];
  for (int i = 0; i < 16; ++i)
    for (int k = 0; k < 32; ++k) hA[i * 32 + k] = (float)((i * 37 + k * 11) % 23 - 11) / 4.0f;
  for (int k = 0; k < 32; ++k)
    for (int j = 0; j < 16; ++j) hB[k * 16 + j] = (float)((k * 13 + j * 7) % 19 - 9) / 4.0f;
  // bf16-rounded reference
  auto r = [](float x) { return (float)(bf16)x; };
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      float s = 0;
      for (int k = 0; k < 32; ++k) s += r(hA[i * 32 + k]) * r(hB[k * 16 + j]);
      ref[i * 16 + j] = s;
    }
  float *dA, *dB, *dD;
  hipMalloc(&dA, sizeof(hA)); hipMalloc(&dB, sizeof(hB)); hipMalloc(&dD, sizeof(hD));
  hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  for (int cand = 0; cand < 5; ++cand) {
    hipMemset(dD, 0, sizeof(hD));
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dD, cand);
    hipMemcpy(hD, dD, sizeof(hD), hipMemcpyDeviceToHost);
    float maxerr = 0;
    for (int x = 0; x < 256; ++x) maxerr = fmaxf(maxerr, fabsf(hD[x] - ref[x]));
    printf("cand %d: maxerr %f %s\n", cand, maxerr, maxerr < 0.15f ? "<== MATCH" : "");
  }
  return 0;
}
