// Tile-shape scan for the bit-permutation tiled permute (k_permute_tile):
// measure GB/s (read+write payload) across (ABITS, BBITS, THREADS)
// variants on a synthetic permutation shaped like the rqc36 step-437 pack
// (2^29 c128 elements, K legs interleaved into A).
//
//   hipcc --offload-arch=gfx950 -O3 scripts/perm_tune.hip -o perm_tune.bin
//   ./perm_tune.bin
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(x)                                    \
  do {                                              \
    hipError_t e = (x);                             \
    if (e != hipSuccess) {                          \
      printf("HIP error %s at %d\n",                \
             hipGetErrorString(e), __LINE__);       \
      exit(1);                                      \
    }                                               \
  } while (0)

typedef unsigned long long u64;
#define MAXBITS 34

struct PermPerm {
  int rbits;
  u64 restS[MAXBITS];
  u64 restD[MAXBITS];
  u64 bS[8];
  u64 bD[8];
  u64 aD[8];
};

__device__ unsigned long long g_bad[4];

template <int AB, int BB, int THREADS>
__global__ __launch_bounds__(THREADS) void k_tile(
    const double2* __restrict__ src, double2* __restrict__ dst, PermPerm pp) {
  constexpr int AN = 1 << AB, BN = 1 << BB;
  __shared__ double2 tile[AN * BN];
  __shared__ u64 sboffS[BN], sboffD[BN], saoffD[AN];
  const int tid = threadIdx.x;
  if (tid < BN) {
    u64 os = 0, od = 0;
    for (int i = 0; i < BB; ++i)
      if (tid >> i & 1) {
        os += pp.bS[i];
        od += pp.bD[i];
      }
    sboffS[tid] = os;
    sboffD[tid] = od;
  } else if (tid < BN + AN) {
    const int a = tid - BN;
    u64 od = 0;
    for (int i = 0; i < AB; ++i)
      if (a >> i & 1) od += pp.aD[i];
    saoffD[a] = od;
  }
  u64 baseS = 0, baseD = 0;
  {
    unsigned r = blockIdx.x;
    for (int i = 0; i < pp.rbits; ++i) {
      if (r & 1) {
        baseS += pp.restS[i];
        baseD += pp.restD[i];
      }
      r >>= 1;
    }
  }
  __syncthreads();
  const u64 LIM = 1ull << 29;
  for (int e = tid; e < AN * BN; e += THREADS) {
    const int a = e & (AN - 1), b = e >> AB;
    u64 idx = baseS + sboffS[b] + (u64)a;
    if (idx >= LIM) {
      g_bad[0] = 1;
      g_bad[1] = idx;
      continue;
    }
    tile[b * AN + ((a ^ b) & (AN - 1))] = src[idx];
  }
  __syncthreads();
  for (int e = tid; e < AN * BN; e += THREADS) {
    const int b = e & (BN - 1), a = e >> BB;
    u64 idx = baseD + saoffD[a] + sboffD[b];
    if (idx >= LIM) {
      g_bad[2] = 1;
      g_bad[3] = idx;
      continue;
    }
    dst[idx] = tile[b * AN + ((a ^ b) & (AN - 1))];
  }
}

int main(int argc, char** argv) {
  const int only = argc > 1 ? atoi(argv[1]) : -1;
  int vidx = 0;
  const int NB = 29;  // 2^29 c128 = 8.6 GB
  const u64 elems = 1ull << NB;
  double2 *src, *dst;
  CHECK(hipMalloc(&src, elems * 16));
  CHECK(hipMalloc(&dst, elems * 16));
  CHECK(hipMemset(src, 1, elems * 16));

  // permutation: dst bit i <- src bit perm[i]; model an interleaved pack
  // (like a_axes reordering): src bits shuffled with stride-2 interleave
  int perm[NB];
  const int shape = argc > 2 ? atoi(argv[2]) : 0;
  if (shape == 0) {  // interleave (rqc36 step-437-like)
    int k = 0;
    for (int i = 0; i * 2 < NB; ++i) perm[k++] = i * 2;
    for (int i = 0; i * 2 + 1 < NB; ++i) perm[k++] = i * 2 + 1;
  } else {  // bit reversal (worst case)
    for (int i = 0; i < NB; ++i) perm[i] = NB - 1 - i;
  }
  // srcStrideOfDstBit[i] = 1 << perm[i]
  auto run = [&](int AB, int BB, int THREADS, auto kern) {
    if (only >= 0 && vidx++ != only) return;
    printf("running AB=%d BB=%d THREADS=%d\n", AB, BB, THREADS);
    fflush(stdout);
    PermPerm pp{};
    // a-bits: dst-low? a-bits must be the SRC-low AB bits (coalesced src):
    // find dst bits whose src stride is 1..1<<(AB-1)
    std::vector<int> dstOfSrc(NB);
    for (int i = 0; i < NB; ++i) dstOfSrc[perm[i]] = i;
    std::vector<char> used(NB, 0);
    for (int i = 0; i < AB; ++i) {  // src bit i -> some dst bit
      pp.aD[i] = 1ull << dstOfSrc[i];
      used[dstOfSrc[i]] = 1;
    }
    int nb = 0;
    for (int d = 0; d < NB && nb < BB; ++d) {  // lowest unused dst bits
      if (used[d]) continue;
      pp.bD[nb] = 1ull << d;
      pp.bS[nb] = 1ull << perm[d];
      used[d] = 1;
      ++nb;
    }
    int nr = 0;
    for (int d = 0; d < NB; ++d) {
      if (used[d]) continue;
      pp.restD[nr] = 1ull << d;
      pp.restS[nr] = 1ull << perm[d];
      ++nr;
    }
    pp.rbits = nr;
    dim3 grid(1u << nr);
    hipEvent_t e0, e1;
    CHECK(hipEventCreate(&e0));
    CHECK(hipEventCreate(&e1));
    kern<<<grid, THREADS>>>(src, dst, pp);  // warmup
    CHECK(hipDeviceSynchronize());
    CHECK(hipEventRecord(e0));
    for (int it = 0; it < 5; ++it) kern<<<grid, THREADS>>>(src, dst, pp);
    CHECK(hipEventRecord(e1));
    CHECK(hipDeviceSynchronize());
    float ms = 0;
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    u64 bad[4] = {};
    CHECK(hipMemcpyFromSymbol(bad, HIP_SYMBOL(g_bad), sizeof bad));
    if (bad[0] || bad[2])
      printf("  BAD: src_oob=%llu (idx %llu) dst_oob=%llu (idx %llu)\n",
             bad[0], bad[1], bad[2], bad[3]);
    double gbs = 5.0 * 2.0 * elems * 16 / (ms / 1e3) / 1e9;
    printf("AB=%d BB=%d THREADS=%4d  %7.1f GB/s (%.2f ms/pass)\n", AB, BB,
           THREADS, gbs, ms / 5.0);
    CHECK(hipEventDestroy(e0));
    CHECK(hipEventDestroy(e1));
  };

  run(6, 6, 512, k_tile<6, 6, 512>);
  run(6, 6, 256, k_tile<6, 6, 256>);
  run(6, 6, 1024, k_tile<6, 6, 1024>);
  run(6, 5, 512, k_tile<6, 5, 512>);
  run(5, 6, 512, k_tile<5, 6, 512>);
  run(5, 5, 512, k_tile<5, 5, 512>);
  run(6, 7, 512, k_tile<6, 7, 512>);  // 128 KB LDS: 1 block/CU
  run(7, 6, 512, k_tile<7, 6, 512>);
  run(6, 5, 1024, k_tile<6, 5, 1024>);
  run(5, 6, 1024, k_tile<5, 6, 1024>);
  run(5, 5, 1024, k_tile<5, 5, 1024>);
  run(6, 4, 512, k_tile<6, 4, 512>);
  return 0;
}
