// Layout probe for v_mfma_scale_f32_32x32x64_f8f6f4 operand fragments.
// Single wave computes one 32x32x64 tile with candidate byte->k maps;
// host compares against a CPU reference to identify the real layout.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

typedef __attribute__((ext_vector_type(8))) int v8i;
typedef __attribute__((ext_vector_type(16))) float v16f;

// candidate L: byte j of lane l maps to k = kmap(l, j)
__device__ int kmap(int cand, int l, int j) {
  switch (cand) {
    case 0: return (l >> 5) * 32 + j;                        // 32 consecutive
    case 1: return 16 * (j >> 3) + (l >> 5) * 8 + (j & 7);   // 4 chained x16
    case 2: return (j >> 3) * 8 + (l >> 5) * 32 + (j & 7);   // blocks of 8? (=c0)
    case 3: return (l >> 5) * 8 + (j >> 3) * 16 + (j & 7);   // alt chain order
    default: return j;
  }
}

__global__ void probe(const char* A, const char* B, float* C, int cand) {
  int l = threadIdx.x;
  unsigned char abuf[32], bbuf[32];
  for (int j = 0; j < 32; ++j) {
    int k = kmap(cand, l, j);
    abuf[j] = A[(l & 31) * 64 + k];
    bbuf[j] = B[(l & 31) * 64 + k];
  }
  v8i a = *(v8i*)abuf;
  v8i b = *(v8i*)bbuf;
  v16f c = {};
  c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
      a, b, c, 0, 0, 0, 0x7F7F7F7Fu, 0, 0x7F7F7F7Fu);
  for (int r = 0; r < 16; ++r) {
    int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    int col = l & 31;
    C[row * 32 + col] = c[r];
  }
}

static float fp8_to_f(unsigned char v) {
  // e4m3fn decode
  int s = v >> 7, e = (v >> 3) & 0xF, m = v & 7;
  if (e == 0xF && m == 7) return nanf("");
  float val;
  if (e == 0) val = ldexpf((float)m / 8.0f, -6);
  else val = ldexpf(1.0f + (float)m / 8.0f, e - 7);
  return s ? -val : val;
}

int main() {
  srand(42);
  unsigned char hA[32 * 64], hB[32 * 64];
  // small integers: bytes for {-2,-1,0,1}
  const unsigned char enc[4] = {0xC0, 0xB8, 0x00, 0x38};
  for (int i = 0; i < 32 * 64; ++i) {
    hA[i] = enc[rand() & 3];
    hB[i] = enc[rand() & 3];
  }
  float ref[32 * 32];
  for (int m = 0; m < 32; ++m)
    for (int n = 0; n < 32; ++n) {
      float acc = 0;
      for (int k = 0; k < 64; ++k)
        acc += fp8_to_f(hA[m * 64 + k]) * fp8_to_f(hB[n * 64 + k]);
      ref[m * 32 + n] = acc;
    }
  char *dA, *dB;
  float* dC;
  (void)hipMalloc(&dA, sizeof(hA));
  (void)hipMalloc(&dB, sizeof(hB));
  (void)hipMalloc(&dC, 32 * 32 * 4);
  (void)hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  for (int cand = 0; cand < 4; ++cand) {
    (void)hipMemset(dC, 0, 32 * 32 * 4);
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dC, cand);
    (void)hipDeviceSynchronize();
    float out[32 * 32];
    (void)hipMemcpy(out, dC, sizeof(out), hipMemcpyDeviceToHost);
    float maxerr = 0;
    for (int i = 0; i < 32 * 32; ++i)
      maxerr = fmaxf(maxerr, fabsf(out[i] - ref[i]));
    printf("cand %d: max_abs_err = %g %s\n", cand, maxerr,
           maxerr == 0 ? "<-- EXACT" : "");
  }
  return 0;
}
