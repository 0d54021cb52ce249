// Standalone repro: f64-MFMA complex GEMM correctness vs CPU for K=64 vs 128.
// Variants probe where the K>64 corruption comes from.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

typedef double v4d __attribute__((ext_vector_type(4)));
#define MF_T 64
#define MF_K 16
#define A_LD 17
#define B_LD 66

// variant 0: the library kernel (copy)
__global__ __launch_bounds__(256) void k0(const double2* A, const double2* B,
                                          double2* C, unsigned long long M,
                                          unsigned long long N,
                                          unsigned long long K) {
  __shared__ double Ar[MF_T * A_LD];
  __shared__ double Ai[MF_T * A_LD];
  __shared__ double Br[MF_K * B_LD];
  __shared__ double Bi[MF_K * B_LD];
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const unsigned long long brow = 0, bcol = 0;
  v4d cr[4], ci[4];
  for (int f = 0; f < 4; ++f) { cr[f] = v4d{0,0,0,0}; ci[f] = v4d{0,0,0,0}; }
  const int fi = lane % 16;
  const int fk = lane / 16;
  for (unsigned long long k0v = 0; k0v < K; k0v += MF_K) {
    for (int i = threadIdx.x; i < MF_T * MF_K; i += 256) {
      int r = i / MF_K, c = i % MF_K;
      double2 v = (r < (int)M && k0v + c < K) ? A[r * K + k0v + c]
                                              : make_double2(0.0, 0.0);
      Ar[r * A_LD + c] = v.x;
      Ai[r * A_LD + c] = v.y;
    }
    for (int i = threadIdx.x; i < MF_K * MF_T; i += 256) {
      int r = i / MF_T, c = i % MF_T;
      double2 v = (k0v + r < K && c < (int)N) ? B[(k0v + r) * N + c]
                                              : make_double2(0.0, 0.0);
      Br[r * B_LD + c] = v.x;
      Bi[r * B_LD + c] = v.y;
    }
    __syncthreads();
    for (int kq = 0; kq < MF_K / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + fk;
      const double ar = Ar[arow * A_LD + ak];
      const double ai = Ai[arow * A_LD + ak];
      for (int f = 0; f < 4; ++f) {
        const int bcolf = f * 16 + fi;
        const double br = Br[ak * B_LD + bcolf];
        const double bi = Bi[ak * B_LD + bcolf];
        cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, br, cr[f], 0, 0, 0);
        cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(-ai, bi, cr[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, bi, ci[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ai, br, ci[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  const int crow0 = wave * 16 + (lane / 16);
  const int ccol = lane % 16;
  for (int f = 0; f < 4; ++f)
    for (int r = 0; r < 4; ++r) {
      unsigned long long row = crow0 + 4 * r, col = f * 16 + ccol;
      if (row < M && col < N) C[row * N + col] = make_double2(cr[f][r], ci[f][r]);
    }
}

// variant 1: kq loop not unrolled
__global__ __launch_bounds__(256) void k1(const double2* A, const double2* B,
                                          double2* C, unsigned long long M,
                                          unsigned long long N,
                                          unsigned long long K) {
  __shared__ double Ar[MF_T * A_LD];
  __shared__ double Ai[MF_T * A_LD];
  __shared__ double Br[MF_K * B_LD];
  __shared__ double Bi[MF_K * B_LD];
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  v4d cr[4], ci[4];
  for (int f = 0; f < 4; ++f) { cr[f] = v4d{0,0,0,0}; ci[f] = v4d{0,0,0,0}; }
  const int fi = lane % 16;
  const int fk = lane / 16;
  for (unsigned long long k0v = 0; k0v < K; k0v += MF_K) {
    for (int i = threadIdx.x; i < MF_T * MF_K; i += 256) {
      int r = i / MF_K, c = i % MF_K;
      double2 v = (r < (int)M && k0v + c < K) ? A[r * K + k0v + c]
                                              : make_double2(0.0, 0.0);
      Ar[r * A_LD + c] = v.x;
      Ai[r * A_LD + c] = v.y;
    }
    for (int i = threadIdx.x; i < MF_K * MF_T; i += 256) {
      int r = i / MF_T, c = i % MF_T;
      double2 v = (k0v + r < K && c < (int)N) ? B[(k0v + r) * N + c]
                                              : make_double2(0.0, 0.0);
      Br[r * B_LD + c] = v.x;
      Bi[r * B_LD + c] = v.y;
    }
    __syncthreads();
#pragma clang loop unroll(disable)
    for (int kq = 0; kq < MF_K / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + fk;
      const double ar = Ar[arow * A_LD + ak];
      const double ai = Ai[arow * A_LD + ak];
#pragma clang loop unroll(disable)
      for (int f = 0; f < 4; ++f) {
        const int bcolf = f * 16 + fi;
        const double br = Br[ak * B_LD + bcolf];
        const double bi = Bi[ak * B_LD + bcolf];
        cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, br, cr[f], 0, 0, 0);
        cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(-ai, bi, cr[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, bi, ci[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ai, br, ci[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  const int crow0 = wave * 16 + (lane / 16);
  const int ccol = lane % 16;
  for (int f = 0; f < 4; ++f)
    for (int r = 0; r < 4; ++r) {
      unsigned long long row = crow0 + 4 * r, col = f * 16 + ccol;
      if (row < M && col < N) C[row * N + col] = make_double2(cr[f][r], ci[f][r]);
    }
}

// variant 2: single accumulator pair (f loop removed; 16x16 out tile per wave,
// grid covers 4 col fragments via blockIdx) — isolates multi-accumulator issue
__global__ __launch_bounds__(256) void k2(const double2* A, const double2* B,
                                          double2* C, unsigned long long M,
                                          unsigned long long N,
                                          unsigned long long K, int f) {
  __shared__ double Ar[MF_T * A_LD];
  __shared__ double Ai[MF_T * A_LD];
  __shared__ double Br[MF_K * B_LD];
  __shared__ double Bi[MF_K * B_LD];
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  v4d cr = v4d{0,0,0,0}, ci = v4d{0,0,0,0};
  const int fi = lane % 16;
  const int fk = lane / 16;
  for (unsigned long long k0v = 0; k0v < K; k0v += MF_K) {
    for (int i = threadIdx.x; i < MF_T * MF_K; i += 256) {
      int r = i / MF_K, c = i % MF_K;
      double2 v = (r < (int)M && k0v + c < K) ? A[r * K + k0v + c]
                                              : make_double2(0.0, 0.0);
      Ar[r * A_LD + c] = v.x;
      Ai[r * A_LD + c] = v.y;
    }
    for (int i = threadIdx.x; i < MF_K * MF_T; i += 256) {
      int r = i / MF_T, c = i % MF_T;
      double2 v = (k0v + r < K && c < (int)N) ? B[(k0v + r) * N + c]
                                              : make_double2(0.0, 0.0);
      Br[r * B_LD + c] = v.x;
      Bi[r * B_LD + c] = v.y;
    }
    __syncthreads();
    for (int kq = 0; kq < MF_K / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + fk;
      const double ar = Ar[arow * A_LD + ak];
      const double ai = Ai[arow * A_LD + ak];
      const int bcolf = f * 16 + fi;
      const double br = Br[ak * B_LD + bcolf];
      const double bi = Bi[ak * B_LD + bcolf];
      cr = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, br, cr, 0, 0, 0);
      cr = __builtin_amdgcn_mfma_f64_16x16x4f64(-ai, bi, cr, 0, 0, 0);
      ci = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, bi, ci, 0, 0, 0);
      ci = __builtin_amdgcn_mfma_f64_16x16x4f64(ai, br, ci, 0, 0, 0);
    }
    __syncthreads();
  }
  const int crow0 = wave * 16 + (lane / 16);
  const int ccol = lane % 16;
  for (int r = 0; r < 4; ++r) {
    unsigned long long row = crow0 + 4 * r, col = f * 16 + ccol;
    if (row < M && col < N) C[row * N + col] = make_double2(cr[r], ci[r]);
  }
}

static void f32_probe();

static void check(const char* name, const std::vector<double2>& got,
                  const std::vector<double2>& ref, int M, int N) {
  double mx = 0; int bad = 0; int first = -1;
  for (int i = 0; i < M * N; ++i) {
    double d = std::abs(got[i].x - ref[i].x) + std::abs(got[i].y - ref[i].y);
    if (d > mx) mx = d;
    if (d > 1e-8) { ++bad; if (first < 0) first = i; }
  }
  printf("%-18s maxdiff=%.3e bad=%d/%d first=(%d,%d)\n", name, mx, bad, M * N,
         first < 0 ? -1 : first / N, first < 0 ? -1 : first % N);
}

int main() {
  for (int K : {64, 128, 256}) {
    int M = 64, N = 64;
    std::vector<double2> A(M * K), B(K * N), C(M * N), R(M * N);
    srand(42);
    auto rnd = []() { return (double)rand() / RAND_MAX - 0.5; };
    for (auto& v : A) { v.x = rnd(); v.y = rnd(); }
    for (auto& v : B) { v.x = rnd(); v.y = rnd(); }
    for (int m = 0; m < M; ++m)
      for (int n = 0; n < N; ++n) {
        double re = 0, im = 0;
        for (int k = 0; k < K; ++k) {
          double2 a = A[m * K + k], b = B[k * N + n];
          re += a.x * b.x - a.y * b.y;
          im += a.x * b.y + a.y * b.x;
        }
        R[m * N + n] = make_double2(re, im);
      }
    double2 *dA, *dB, *dC;
    hipMalloc(&dA, A.size() * 16); hipMalloc(&dB, B.size() * 16);
    hipMalloc(&dC, C.size() * 16);
    hipMemcpy(dA, A.data(), A.size() * 16, hipMemcpyHostToDevice);
    hipMemcpy(dB, B.data(), B.size() * 16, hipMemcpyHostToDevice);
    printf("--- K=%d ---\n", K);
    hipMemset(dC, 0, C.size() * 16);
    hipLaunchKernelGGL(k0, dim3(1), dim3(256), 0, 0, dA, dB, dC, M, N, K);
    hipMemcpy(C.data(), dC, C.size() * 16, hipMemcpyDeviceToHost);
    check("k0 (library)", C, R, M, N);
    hipMemset(dC, 0, C.size() * 16);
    hipLaunchKernelGGL(k1, dim3(1), dim3(256), 0, 0, dA, dB, dC, M, N, K);
    hipMemcpy(C.data(), dC, C.size() * 16, hipMemcpyDeviceToHost);
    check("k1 (no-unroll)", C, R, M, N);
    hipMemset(dC, 0, C.size() * 16);
    for (int f = 0; f < 4; ++f)
      hipLaunchKernelGGL(k2, dim3(1), dim3(256), 0, 0, dA, dB, dC, M, N, K, f);
    hipMemcpy(C.data(), dC, C.size() * 16, hipMemcpyDeviceToHost);
    check("k2 (1-acc x4)", C, R, M, N);
    hipFree(dA); hipFree(dB); hipFree(dC);
  }
  f32_probe();
  return 0;
}

// f32 16x16x4 D-layout probe: computes one MFMA with asymmetric operands
// and reports which (lane, reg) -> row mapping matches the CPU result.
typedef float v4f __attribute__((ext_vector_type(4)));
__global__ void k_f32probe(float* D) {
  int l = threadIdx.x;
  float a = (float)((l % 16) * 8 + (l / 16));       // A[i][k] = i*8+k
  float b = (float)((l / 16) * 100 + (l % 16) * 3); // B[k][j] = 100k+3j
  v4f acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) D[l * 4 + r] = acc[r];
}

static void f32_probe() {
  float* dD;
  (void)hipMalloc(&dD, 64 * 4 * sizeof(float));
  hipLaunchKernelGGL(k_f32probe, dim3(1), dim3(64), 0, 0, dD);
  std::vector<float> D(256);
  (void)hipMemcpy(D.data(), dD, 256 * 4, hipMemcpyDeviceToHost);
  // CPU reference C[i][j] = sum_k A[i][k]*B[k][j]
  float C[16][16];
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      float s = 0;
      for (int k = 0; k < 4; ++k) s += (i * 8 + k) * (100.f * k + 3.f * j);
      C[i][j] = s;
    }
  int ok_a = 0, ok_b = 0;
  for (int l = 0; l < 64; ++l)
    for (int r = 0; r < 4; ++r) {
      int col = l % 16;
      if (D[l * 4 + r] == C[(l / 16) * 4 + r][col]) ++ok_a;  // row=(l/16)*4+r
      if (D[l * 4 + r] == C[4 * r + l / 16][col]) ++ok_b;    // row=4r+l/16
    }
  printf("f32 16x16x4 D-map: (l/16)*4+r matches %d/256; 4r+l/16 matches %d/256\n",
         ok_a, ok_b);
}
