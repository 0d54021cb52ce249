// f64 MFMA peak microbenchmark: v_mfma_f64_16x16x4_f64 back-to-back on 8
// independent accumulators, one wave per SIMD. Pins the `peak` used for
// bench.py's roofline (spec claim: 78.6 TF/s at 2.4 GHz).
#include <hip/hip_runtime.h>
#include <cstdio>

typedef double v4d __attribute__((ext_vector_type(4)));

template <int ACCS>
__global__ __launch_bounds__(256) void peak_kernel(double* out, int iters) {
  v4d acc[ACCS];
  for (int i = 0; i < ACCS; ++i) acc[i] = v4d{0.0, 0.0, 0.0, 0.0};
  double a = 1.0 + threadIdx.x * 1e-9;
  double b = 1.0 - threadIdx.x * 1e-9;
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < ACCS; ++i)
      acc[i] = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc[i], 0, 0, 0);
  }
  double s = 0;
  for (int i = 0; i < ACCS; ++i)
    s += acc[i][0] + acc[i][1] + acc[i][2] + acc[i][3];
  if (s == -1.0) out[threadIdx.x] = s;  // never true; defeats DCE
}

template <int ACCS>
static void run(int blocks, int iters, double* out) {
  dim3 grid(blocks), block(256);
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  float best = 1e30f;
  for (int rep = 0; rep < 3; ++rep) {
    (void)hipEventRecord(e0);
    hipLaunchKernelGGL(peak_kernel<ACCS>, grid, block, 0, 0, out, iters);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms;
    (void)hipEventElapsedTime(&ms, e0, e1);
    if (ms < best) best = ms;
  }
  double flops =
      2048.0 * ACCS * (double)iters * (blocks * (double)256 / 64.0);
  printf("blocks/CU=%d accs=%d: %.2f TF/s\n", blocks / 256, ACCS,
         flops / (best / 1e3) / 1e12);
}

int main() {
  double* out;
  (void)hipMalloc(&out, 256 * 8);
  (void)hipDeviceSynchronize();
  for (int bpc = 1; bpc <= 4; bpc *= 2) {
    run<4>(bpc * 256, 20000 / bpc, out);
    run<8>(bpc * 256, 20000 / bpc, out);
    run<16>(bpc * 256, 10000 / bpc, out);
  }
  return 0;
}
