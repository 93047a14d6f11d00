// Empirical determination of the A/B fragment layouts of
// __builtin_amdgcn_mfma_f32_16x16x32_bf16 on gfx950.
//
// Known (guide §3, measured): C/D layout is col=lane&15,
// row=(lane>>4)*4+reg.  A/B per-lane element count is 8 bf16.
// This probe tests the candidate lane->(row,k) mappings against a CPU
// reference with random asymmetric A, B and prints which matches.
//
// Build: hipcc --offload-arch=gfx950 -O2 tools/mfma_probe.hip -o /tmp/mfma_probe
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

// hypothesis for lane->(outer,k) mapping of an 8-element fragment
// H=0: contiguous: k = (lane>>4)*8 + j
// H=1: split:      k = (lane>>4)*4 + (j&3) + (j>>2)*16
__device__ __forceinline__ int frag_k(int lane, int j, int H) {
  if (H == 0) return (lane >> 4) * 8 + j;
  return (lane >> 4) * 4 + (j & 3) + (j >> 2) * 16;
}

template <int HA, int HB>
__global__ void probe_kernel(const float* A, const float* B, float* D) {
  // A: 16x32 row-major, B: 32x16 row-major, D: 16x16 row-major
  int lane = threadIdx.x & 63;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    int ka = frag_k(lane, j, HA);
    int kb = frag_k(lane, j, HB);
    a[j] = (__bf16)A[(lane & 15) * 32 + ka];   // A[row][k]
    b[j] = (__bf16)B[kb * 16 + (lane & 15)];   // B[k][col]
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  for (int reg = 0; reg < 4; ++reg) {
    int row = (lane >> 4) * 4 + reg;
    int col = lane & 15;
    D[row * 16 + col] = c[reg];
  }
}

int main() {
  std::vector<float> A(16 * 32), B(32 * 16), Dref(16 * 16);
  srand(7);
  auto rnd = [] { return (float)((rand() % 17) - 8); };  // bf16-exact ints
  for (auto& v : A) v = rnd();
  for (auto& v : B) v = rnd();
  for (int r = 0; r < 16; ++r)
    for (int c = 0; c < 16; ++c) {
      float acc = 0;
      for (int k = 0; k < 32; ++k) acc += A[r * 32 + k] * B[k * 16 + c];
      Dref[r * 16 + c] = acc;
    }

  float *dA, *dB, *dD;
  hipMalloc(&dA, A.size() * 4);
  hipMalloc(&dB, B.size() * 4);
  hipMalloc(&dD, Dref.size() * 4);
  hipMemcpy(dA, A.data(), A.size() * 4, hipMemcpyHostToDevice);
  hipMemcpy(dB, B.data(), B.size() * 4, hipMemcpyHostToDevice);

  std::vector<float> D(16 * 16);
  auto check = [&](const char* name) {
    hipMemcpy(D.data(), dD, D.size() * 4, hipMemcpyDeviceToHost);
    int bad = 0;
    for (int i = 0; i < 256; ++i)
      if (fabsf(D[i] - Dref[i]) > 0.5f) bad++;
    printf("%s: %s (%d/256 mismatches)\n", name,
           bad == 0 ? "MATCH" : "no", bad);
    return bad == 0;
  };

#define RUN(HA, HB)                                                     \
  hipLaunchKernelGGL((probe_kernel<HA, HB>), dim3(1), dim3(64), 0, 0,   \
                     dA, dB, dD);                                       \
  hipDeviceSynchronize();                                               \
  check("A-H" #HA " B-H" #HB);

  RUN(0, 0);
  RUN(0, 1);
  RUN(1, 0);
  RUN(1, 1);
  return 0;
}
