// f64 MFMA peak microbenchmark: v_mfma_f64_16x16x4_f64 back-to-back on 8
// independent accumulators, one wave per SIMD. Pins the `peak` used for
// bench.py's roofline (spec claim: 78.6 TF/s at 2.4 GHz).
#include <hip/hip_runtime.h>
#include <cstdio>

typedef double v4d __attribute__((ext_vector_type(4)));

__global__ __launch_bounds__(256) void peak_kernel(double* out, int iters) {
  v4d acc[8];
  for (int i = 0; i < 8; ++i) acc[i] = v4d{0.0, 0.0, 0.0, 0.0};
  double a = 1.0 + threadIdx.x * 1e-9;
  double b = 1.0 - threadIdx.x * 1e-9;
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i] = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc[i], 0, 0, 0);
  }
  double s = 0;
  for (int i = 0; i < 8; ++i) s += acc[i][0] + acc[i][1] + acc[i][2] + acc[i][3];
  if (s == -1.0) out[threadIdx.x] = s;  // never true; defeats DCE
}

int main() {
  int iters = 20000;
  double* out;
  (void)hipMalloc(&out, 256 * 8);
  // grid: 4 waves per block, 256 blocks x 4 = 1024 waves = 1 per SIMD
  dim3 grid(256), block(256);
  (void)hipDeviceSynchronize();
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  for (int rep = 0; rep < 3; ++rep) {
    (void)hipEventRecord(e0);
    hipLaunchKernelGGL(peak_kernel, grid, block, 0, 0, out, iters);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms;
    (void)hipEventElapsedTime(&ms, e0, e1);
    // flops: 16*16*4*2 = 2048 per MFMA, 8 per iter per wave, 1024 waves
    double flops = 2048.0 * 8.0 * iters * (grid.x * (double)block.x / 64.0);
    printf("f64 MFMA peak rep%d: %.2f TF/s (%.3f ms)\n", rep,
           flops / (ms / 1e3) / 1e12, ms);
  }
  return 0;
}
