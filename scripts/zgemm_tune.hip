// zgemm variant tuning harness: times templated variants of the c128 MFMA
// GEMM on a given (M, N, K), checks each against the baseline.
// Usage: zgemm_tune [M N K]
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

typedef double v4d __attribute__((ext_vector_type(4)));
typedef unsigned long long u64;

// Templated kernel: TM rows (= WAVES*16), 64 cols, KT-deep K tile.
// REORDER: 0 = per-fragment 4-MFMA chain, 1 = pass-per-term (dep distance 4).
template <int WAVES, int KT, int REORDER, int FRAGS = 4, int GROUPC = 0>
__global__ __launch_bounds__(WAVES * 64) void zg(
    const double2* __restrict__ A, const double2* __restrict__ B,
    double2* __restrict__ C, u64 M, u64 N, u64 K, u64 col_tiles) {
  constexpr int TM = WAVES * 16;
  constexpr int TN = FRAGS * 16;
  constexpr int ALD = KT + 1;
  constexpr int BLD = TN + 2;
  __shared__ double Ar[TM * ALD];
  __shared__ double Ai[TM * ALD];
  __shared__ double Br[KT * BLD];
  __shared__ double Bi[KT * BLD];
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const u64 tile = blockIdx.x;
  u64 trow, tcol;
  if (GROUPC > 0) {
    const u64 row_tiles = gridDim.x / col_tiles;
    const u64 g = tile / (row_tiles * GROUPC);
    const u64 rem = tile - g * row_tiles * GROUPC;
    const u64 wc = (GROUPC < col_tiles - g * GROUPC) ? GROUPC
                                                     : col_tiles - g * GROUPC;
    trow = rem / wc;
    tcol = g * GROUPC + rem % wc;
  } else {
    trow = tile / col_tiles;
    tcol = tile % col_tiles;
  }
  const u64 brow = trow * TM, bcol = tcol * TN;
  v4d cr[FRAGS], ci[FRAGS];
  v4d p3[REORDER == 2 ? FRAGS : 1];
  for (int f = 0; f < FRAGS; ++f) {
    cr[f] = v4d{0, 0, 0, 0};
    ci[f] = v4d{0, 0, 0, 0};
  }
  if (REORDER == 2)
    for (int f = 0; f < FRAGS; ++f) p3[f] = v4d{0, 0, 0, 0};
  const int fi = lane % 16;
  const int fk = lane / 16;
  const int NT = WAVES * 64;
  for (u64 k0 = 0; k0 < K; k0 += KT) {
    for (int i = threadIdx.x; i < TM * KT; i += NT) {
      int r = i / KT, c = i % KT;
      double2 v = (brow + r < M && k0 + c < K) ? A[(brow + r) * K + k0 + c]
                                               : make_double2(0.0, 0.0);
      Ar[r * ALD + c] = v.x;
      Ai[r * ALD + c] = v.y;
    }
    for (int i = threadIdx.x; i < KT * TN; i += NT) {
      int r = i / TN, c = i % TN;
      double2 v = (k0 + r < K && bcol + c < N) ? B[(k0 + r) * N + bcol + c]
                                               : make_double2(0.0, 0.0);
      Br[r * BLD + c] = v.x;
      Bi[r * BLD + c] = v.y;
    }
    __syncthreads();
    for (int kq = 0; kq < KT / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + fk;
      const double ar = Ar[arow * ALD + ak];
      const double ai = Ai[arow * ALD + ak];
      double br[FRAGS], bi[FRAGS];
      for (int f = 0; f < FRAGS; ++f) {
        br[f] = Br[ak * BLD + f * 16 + fi];
        bi[f] = Bi[ak * BLD + f * 16 + fi];
      }
      if (REORDER == 2) {
        // Gauss 3-mult: cr accumulates ArBr, ci accumulates AiBi,
        // p3 accumulates (Ar+Ai)(Br+Bi); combine in the epilogue.
        const double as = ar + ai;
        for (int f = 0; f < FRAGS; ++f) {
          cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, br[f], cr[f], 0, 0, 0);
          ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ai, bi[f], ci[f], 0, 0, 0);
          p3[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(as, br[f] + bi[f], p3[f], 0, 0, 0);
        }
      } else if (REORDER == 0) {
        for (int f = 0; f < FRAGS; ++f) {
          cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, br[f], cr[f], 0, 0, 0);
          cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(-ai, bi[f], cr[f], 0, 0, 0);
          ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, bi[f], ci[f], 0, 0, 0);
          ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ai, br[f], ci[f], 0, 0, 0);
        }
      } else {
        for (int f = 0; f < FRAGS; ++f)
          cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, br[f], cr[f], 0, 0, 0);
        for (int f = 0; f < FRAGS; ++f)
          ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, bi[f], ci[f], 0, 0, 0);
        for (int f = 0; f < FRAGS; ++f)
          cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(-ai, bi[f], cr[f], 0, 0, 0);
        for (int f = 0; f < FRAGS; ++f)
          ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(ai, br[f], ci[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  const int crow0 = wave * 16 + (lane / 16);
  const int ccol = lane % 16;
  for (int f = 0; f < FRAGS; ++f)
    for (int r = 0; r < 4; ++r) {
      u64 row = brow + crow0 + 4 * r;
      u64 col = bcol + f * 16 + ccol;
      if (row >= M || col >= N) continue;
      if (REORDER == 2)
        C[row * N + col] = make_double2(
            cr[f][r] - ci[f][r], p3[f][r] - cr[f][r] - ci[f][r]);
      else
        C[row * N + col] = make_double2(cr[f][r], ci[f][r]);
    }
}


// glds variant: LDS-DMA staging (no staging registers / ds_writes),
// interleaved c128 LDS image with XOR swizzle, operands read as
// ds_read_b128 (re+im in one instruction). 128x64 tile, 8 waves, KT=16.
__global__ __launch_bounds__(512) void zg_glds(
    const double2* __restrict__ A, const double2* __restrict__ B,
    double2* __restrict__ C, u64 M, u64 N, u64 K, u64 col_tiles) {
  constexpr int TM = 128, TN = 64, KT = 16;
  __shared__ double2 As[TM * KT];  // [r][c ^ (r & 15)]
  __shared__ double2 Bs[KT * TN];  // [k][j ^ ((k & 3) << 4)]
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const u64 tile = blockIdx.x;
  const u64 brow = (tile / col_tiles) * TM, bcol = (tile % col_tiles) * TN;
  v4d cr[4], ci[4];
  for (int f = 0; f < 4; ++f) { cr[f] = v4d{0,0,0,0}; ci[f] = v4d{0,0,0,0}; }
  const int fi = lane % 16;
  const int fk = lane / 16;
  // glds lane mapping: each wave-wide glds writes 64 consecutive 16B LDS
  // slots starting at a wave-uniform base; lane l supplies the element that
  // belongs at base + l. A tile: 128*16 = 2048 slots = 4 glds per wave
  // (8 waves): wave w, piece p covers rows [ (w*4+p)*... ]. Linear LDS index
  // i = r*KT + c_sw; we choose src so that image[i] = A[r][c] with
  // c = c_sw ^ (r & 15).
  for (u64 k0 = 0; k0 < K; k0 += KT) {
    for (int piece = 0; piece < 4; ++piece) {
      // LDS dst = wave-uniform base + lane*16 (implicit); choose the SOURCE
      // per lane so the swizzled image lands linearly.
      int base = (wave * 4 + piece) * 64;
      int i = base + lane;
      int r = i / KT, c_sw = i % KT;
      int c = c_sw ^ (r & 15);
      const double2* src = &A[(brow + r) * K + k0 + c];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)&As[base], 16, 0, 0);
    }
    for (int piece = 0; piece < 2; ++piece) {
      int base = piece * 512 + wave * 64;
      int j = base + lane;
      int k = j / TN, col_sw = j % TN;
      int col = col_sw ^ ((k & 3) << 4);
      const double2* src = &B[(k0 + k) * N + bcol + col];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)&Bs[base], 16, 0, 0);
    }
    __syncthreads();  // carries vmcnt(0): drains the DMAs
    for (int kq = 0; kq < KT / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + fk;
      double2 a = As[arow * KT + (ak ^ (arow & 15))];
      for (int f = 0; f < 4; ++f) {
        const int bcolf = f * 16 + fi;
        double2 b = Bs[ak * TN + (bcolf ^ ((ak & 3) << 4))];
        cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(a.x, b.x, cr[f], 0, 0, 0);
        cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(-a.y, b.y, cr[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(a.x, b.y, ci[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(a.y, b.x, ci[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  const int crow0 = wave * 16 + (lane / 16);
  const int ccol = lane % 16;
  for (int f = 0; f < 4; ++f)
    for (int r = 0; r < 4; ++r) {
      u64 row = brow + crow0 + 4 * r;
      u64 col = bcol + f * 16 + ccol;
      if (row < M && col < N) C[row * N + col] = make_double2(cr[f][r], ci[f][r]);
    }
}

static void launch_glds(const double2* A, const double2* B, double2* C, u64 M,
                        u64 N, u64 K) {
  u64 rt = (M + 127) / 128, ct = (N + 63) / 64;
  hipLaunchKernelGGL(zg_glds, dim3((unsigned)(rt * ct)), dim3(512), 0, 0, A,
                     B, C, M, N, K, ct);
}


// glds double-buffered: 2 LDS buffers, raw barrier, counted vmcnt — the
// next K-tile's DMAs overlap the current tile's MFMAs. Wave w's A slots are
// self-written (rows 16w..16w+16); B slots are cross-wave (barrier covers).
__global__ __launch_bounds__(512) void zg_glds_db(
    const double2* __restrict__ A, const double2* __restrict__ B,
    double2* __restrict__ C, u64 M, u64 N, u64 K, u64 col_tiles) {
  constexpr int TM = 128, TN = 64, KT = 16;
  constexpr int ASLOTS = TM * KT, BSLOTS = KT * TN;
  __shared__ double2 lds[2 * (ASLOTS + BSLOTS)];
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const u64 tile = blockIdx.x;
  const u64 brow = (tile / col_tiles) * TM, bcol = (tile % col_tiles) * TN;
  v4d cr[4], ci[4];
  for (int f = 0; f < 4; ++f) { cr[f] = v4d{0,0,0,0}; ci[f] = v4d{0,0,0,0}; }
  const int fi = lane % 16;
  const int fk = lane / 16;

  auto stage = [&](int buf, u64 k0) {
    double2* As = lds + buf * (ASLOTS + BSLOTS);
    double2* Bs = As + ASLOTS;
    for (int piece = 0; piece < 4; ++piece) {
      int base = (wave * 4 + piece) * 64;
      int i = base + lane;
      int r = i / KT, c_sw = i % KT;
      int c = c_sw ^ (r & 15);
      const double2* src = &A[(brow + r) * K + k0 + c];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)&As[base], 16, 0, 0);
    }
    for (int piece = 0; piece < 2; ++piece) {
      int base = piece * 512 + wave * 64;
      int j = base + lane;
      int k = j / TN, col_sw = j % TN;
      int col = col_sw ^ ((k & 3) << 4);
      const double2* src = &B[(k0 + k) * N + bcol + col];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)&Bs[base], 16, 0, 0);
    }
  };

  stage(0, 0);
  int buf = 0;
  for (u64 k0 = 0; k0 < K; k0 += KT) {
    if (k0 + KT < K) stage(buf ^ 1, k0 + KT);
    // wait for the CURRENT buffer's own 6 DMAs (the 6 just issued for the
    // next buffer stay in flight), then sync all waves (B is cross-wave)
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    const double2* As = lds + buf * (ASLOTS + BSLOTS);
    const double2* Bs = As + ASLOTS;
    for (int kq = 0; kq < KT / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + fk;
      double2 a = As[arow * KT + (ak ^ (arow & 15))];
      for (int f = 0; f < 4; ++f) {
        const int bcolf = f * 16 + fi;
        double2 b = Bs[ak * TN + (bcolf ^ ((ak & 3) << 4))];
        cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(a.x, b.x, cr[f], 0, 0, 0);
        cr[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(-a.y, b.y, cr[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(a.x, b.y, ci[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f64_16x16x4f64(a.y, b.x, ci[f], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }
  const int crow0 = wave * 16 + (lane / 16);
  const int ccol = lane % 16;
  for (int f = 0; f < 4; ++f)
    for (int r = 0; r < 4; ++r) {
      u64 row = brow + crow0 + 4 * r;
      u64 col = bcol + f * 16 + ccol;
      if (row < M && col < N) C[row * N + col] = make_double2(cr[f][r], ci[f][r]);
    }
}

static void launch_glds_db(const double2* A, const double2* B, double2* C,
                           u64 M, u64 N, u64 K) {
  u64 rt = (M + 127) / 128, ct = (N + 63) / 64;
  hipLaunchKernelGGL(zg_glds_db, dim3((unsigned)(rt * ct)), dim3(512), 0, 0,
                     A, B, C, M, N, K, ct);
}

struct Variant {
  const char* name;
  void (*launch)(const double2*, const double2*, double2*, u64, u64, u64);
};

template <int WAVES, int KT, int REORDER, int FRAGS = 4, int GROUPC = 0>
static void launch_zg(const double2* A, const double2* B, double2* C, u64 M,
                      u64 N, u64 K) {
  constexpr int TM = WAVES * 16;
  constexpr int TN = FRAGS * 16;
  u64 rt = (M + TM - 1) / TM, ct = (N + TN - 1) / TN;
  hipLaunchKernelGGL((zg<WAVES, KT, REORDER, FRAGS, GROUPC>),
                     dim3((unsigned)(rt * ct)), dim3(WAVES * 64), 0, 0, A, B,
                     C, M, N, K, ct);
}

int main(int argc, char** argv) {
  u64 M = argc > 1 ? atoll(argv[1]) : 16384;
  u64 N = argc > 2 ? atoll(argv[2]) : 8192;
  u64 K = argc > 3 ? atoll(argv[3]) : 8192;
  std::vector<double2> hA(1), hB(1);
  double2 *A, *B, *C, *Cref;
  (void)hipMalloc(&A, M * K * 16);
  (void)hipMalloc(&B, K * N * 16);
  (void)hipMalloc(&C, M * N * 16);
  (void)hipMalloc(&Cref, M * N * 16);
  // fill with pseudo-random device-side pattern via a tiny kernel
  auto fill = [](double2* p, u64 n, int seed) {
    struct L {
      static __global__ void f(double2* p, u64 n, int seed) {
        for (u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x; i < n;
             i += gridDim.x * (u64)blockDim.x) {
          // FULL-entropy mantissas: low-entropy values flatter DVFS-bound
          // kernels by ~20% (MI355X_MICROARCH.md "DVFS give-back")
          unsigned x = (unsigned)(i * 2654435761u) ^ (seed * 40503u);
          x ^= x >> 13;
          x *= 0x5bd1e995u;
          x ^= x >> 15;
          unsigned y = x * 1664525u + 1013904223u;
          unsigned z = y * 22695477u + 1u;
          p[i] = make_double2(
              ((double)x + (double)y * 2.3283064365386963e-10) /
                      4294967296.0 - 0.5,
              ((double)y + (double)z * 2.3283064365386963e-10) /
                      4294967296.0 - 0.5);
        }
      }
    };
    hipLaunchKernelGGL(L::f, dim3(4096), dim3(256), 0, 0, p, n, seed);
  };
  fill(A, M * K, 1);
  fill(B, K * N, 2);
  (void)hipDeviceSynchronize();

  Variant variants[] = {
      {"w8 k16 (base)", launch_zg<8, 16, 0>},
      {"w8 k16 glds", launch_glds},
      {"w8 k16 glds-db", launch_glds_db},
  };
  double flops = 8.0 * M * N * K;
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  bool first = true;
  for (auto& v : variants) {
    v.launch(A, B, first ? Cref : C, M, N, K);  // warm + reference
    (void)hipDeviceSynchronize();
    float best = 1e30f;
    for (int rep = 0; rep < 3; ++rep) {
      (void)hipEventRecord(e0);
      v.launch(A, B, first ? Cref : C, M, N, K);
      (void)hipEventRecord(e1);
      (void)hipEventSynchronize(e1);
      float ms;
      (void)hipEventElapsedTime(&ms, e0, e1);
      if (ms < best) best = ms;
    }
    double tf = flops / (best / 1e3) / 1e12;
    double err = 0.0;
    if (!first) {
      // device-side compare (sampled): copy a strip
      u64 sample = M * N > (u64)1 << 22 ? (u64)1 << 22 : M * N;
      std::vector<double2> g(sample), r(sample);
      (void)hipMemcpy(g.data(), C, sample * 16, hipMemcpyDeviceToHost);
      (void)hipMemcpy(r.data(), Cref, sample * 16, hipMemcpyDeviceToHost);
      for (u64 i = 0; i < sample; ++i)
        err = fmax(err, fabs(g[i].x - r[i].x) + fabs(g[i].y - r[i].y));
    }
    printf("%-20s %8.3f ms  %7.2f TF/s  maxdiff_vs_v0=%.2e\n", v.name, best,
           tf, err);
    first = false;
  }
  return 0;
}
