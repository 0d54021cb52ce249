// tnc_hip — MI355X-native (gfx950/CDNA4) tensor-network contraction library.
//
// Implements the C ABI of include/tnc_hip.h: pairwise complex128 einsum
// (the tblis::tensor_mult replacement, reference call site
// tnc/src/tensornetwork/contraction.rs:111-113) as hand-written HIP kernels,
// and the device-resident replace-left network executor (the
// contract_tensor_network replacement, contraction.rs:35-68).
//
// Kernel classes (DESIGN.md "Data layout & kernels"):
//   smallk — K <= 64: one thread per output element, K-offsets precomputed
//            in LDS, gathered strided reads, coalesced writes. Covers gate
//            applications (K,N tiny, M huge) and outer products (K == 1).
//   anyk   — skinny shapes with K > 64: same, offsets computed inline.
//   dot    — M == N == 1, large K: two-pass block reduction.
//   zgemm  — TTGT: pack (index permute, skipped when the operand is already
//            contiguous in GEMM order) + tiled complex GEMM
//            (v_mfma_f64_16x16x4_f64 for large tiles, LDS-tiled VALU
//            fallback for ragged shapes) + optional unpack permute.
//
// All index maps have a pow2 fast path (every bond dim is 2 in the benchmark
// configs => index maps are pure bit shuffles) and a general div/mod path
// (the reference's golden-vector tensors use dims 3..8).

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <string>
#include <type_traits>
#include <vector>

#include "../../include/tnc_hip.h"

typedef uint64_t u64;
typedef int64_t i64;

// ---------------------------------------------------------------------------
// error plumbing
// ---------------------------------------------------------------------------

static thread_local std::string g_last_error;

extern "C" const char* tn_last_error(void) { return g_last_error.c_str(); }

#define FAILV(code, ...)                       \
  do {                                         \
    char buf_[512];                            \
    snprintf(buf_, sizeof(buf_), __VA_ARGS__); \
    g_last_error = buf_;                       \
    return (code);                             \
  } while (0)

#define HIP_CHECK(expr)                                                 \
  do {                                                                  \
    hipError_t err_ = (expr);                                           \
    if (err_ != hipSuccess)                                             \
      FAILV(TN_ERR_HIP, "%s failed: %s (%s:%d)", #expr,                 \
            hipGetErrorString(err_), __FILE__, __LINE__);               \
  } while (0)

// ---------------------------------------------------------------------------
// gather maps
// ---------------------------------------------------------------------------

#define TN_MAXR 40   // max axes per map
#define TN_SMALLK 64 // LDS-precomputed K offsets

struct GatherMap {
  int n;
  int pow2;             // if set: pstride holds shifts, dim holds masks
  u64 pstride[TN_MAXR]; // packed-linear stride (suffix product) or shift
  u64 dim[TN_MAXR];     // axis dim, or mask (dim-1) when pow2
  i64 sa[TN_MAXR];      // source-A stride in elements (0 if absent)
  i64 sb[TN_MAXR];      // source-B stride in elements
};

template <bool P2>
__device__ __forceinline__ void gather2(const GatherMap& m, u64 p, i64& oa,
                                        i64& ob) {
  i64 a = 0, b = 0;
  for (int i = 0; i < m.n; ++i) {
    u64 c = P2 ? ((p >> m.pstride[i]) & m.dim[i])
               : ((p / m.pstride[i]) % m.dim[i]);
    a += (i64)c * m.sa[i];
    b += (i64)c * m.sb[i];
  }
  oa = a;
  ob = b;
}

template <bool P2>
__device__ __forceinline__ i64 gather1(const GatherMap& m, u64 p) {
  i64 a = 0;
  for (int i = 0; i < m.n; ++i) {
    u64 c = P2 ? ((p >> m.pstride[i]) & m.dim[i])
               : ((p / m.pstride[i]) % m.dim[i]);
    a += (i64)c * m.sa[i];
  }
  return a;
}

// ---------------------------------------------------------------------------
// kernels
// ---------------------------------------------------------------------------

template <bool P2, typename CT>
__global__ void k_einsum_smallk(const CT* __restrict__ A,
                                const CT* __restrict__ B,
                                CT* __restrict__ C, u64 nout,
                                GatherMap omap, GatherMap kmap, int K) {
  using RT = decltype(CT{}.x);
  __shared__ i64 koffA[TN_SMALLK];
  __shared__ i64 koffB[TN_SMALLK];
  if (threadIdx.x < (unsigned)K) {
    i64 ka, kb;
    gather2<P2>(kmap, threadIdx.x, ka, kb);
    koffA[threadIdx.x] = ka;
    koffB[threadIdx.x] = kb;
  }
  __syncthreads();
  for (u64 p = blockIdx.x * (u64)blockDim.x + threadIdx.x; p < nout;
       p += gridDim.x * (u64)blockDim.x) {
    i64 oa, ob;
    gather2<P2>(omap, p, oa, ob);
    RT re = 0, im = 0;
    for (int k = 0; k < K; ++k) {
      CT a = A[oa + koffA[k]];
      CT b = B[ob + koffB[k]];
      re = fma(a.x, b.x, fma(-a.y, b.y, re));
      im = fma(a.x, b.y, fma(a.y, b.x, im));
    }
    C[p] = CT{re, im};
  }
}

// Table-decode smallk: for pow2 maps the source offset is additive over the
// index bits, so oa(p)/ob(p) collapse to four 8-bit LDS lookups instead of
// an n-axis shift/mask loop per element. The decode, not memory, bounds
// k_einsum_smallk on rank-~30 intermediates (measured 0.8 TB/s effective on
// a 2^30-element gate-apply step). Requires nout < 2^32.
template <typename CT>
__global__ void k_einsum_smallk_tbl(const CT* __restrict__ A,
                                    const CT* __restrict__ B,
                                    CT* __restrict__ C, u64 nout,
                                    GatherMap omap, GatherMap kmap, int K) {
  using RT = decltype(CT{}.x);
  __shared__ i64 ta[4][256], tb[4][256];
  __shared__ i64 koffA[TN_SMALLK];
  __shared__ i64 koffB[TN_SMALLK];
  for (int t = threadIdx.x; t < 1024; t += blockDim.x) {
    const int ti = t >> 8, e = t & 255;
    i64 oa, ob;
    gather2<true>(omap, (u64)e << (8 * ti), oa, ob);
    ta[ti][e] = oa;
    tb[ti][e] = ob;
  }
  if (threadIdx.x < (unsigned)K) {
    i64 ka, kb;
    gather2<true>(kmap, threadIdx.x, ka, kb);
    koffA[threadIdx.x] = ka;
    koffB[threadIdx.x] = kb;
  }
  __syncthreads();
  for (u64 p = blockIdx.x * (u64)blockDim.x + threadIdx.x; p < nout;
       p += gridDim.x * (u64)blockDim.x) {
    const unsigned lo = (unsigned)p;
    const i64 oa = ta[0][lo & 255] + ta[1][(lo >> 8) & 255] +
                   ta[2][(lo >> 16) & 255] + ta[3][lo >> 24];
    const i64 ob = tb[0][lo & 255] + tb[1][(lo >> 8) & 255] +
                   tb[2][(lo >> 16) & 255] + tb[3][lo >> 24];
    RT re = 0, im = 0;
    for (int k = 0; k < K; ++k) {
      CT a = A[oa + koffA[k]];
      CT b = B[ob + koffB[k]];
      re = fma(a.x, b.x, fma(-a.y, b.y, re));
      im = fma(a.x, b.y, fma(a.y, b.x, im));
    }
    C[p] = CT{re, im};
  }
}

// skinny shapes with K > TN_SMALLK: offsets computed inline per k.
template <bool P2, typename CT>
__global__ void k_einsum_anyk(const CT* __restrict__ A,
                              const CT* __restrict__ B,
                              CT* __restrict__ C, u64 nout, GatherMap omap,
                              GatherMap kmap, u64 K) {
  using RT = decltype(CT{}.x);
  for (u64 p = blockIdx.x * (u64)blockDim.x + threadIdx.x; p < nout;
       p += gridDim.x * (u64)blockDim.x) {
    i64 oa, ob;
    gather2<P2>(omap, p, oa, ob);
    RT re = 0, im = 0;
    for (u64 k = 0; k < K; ++k) {
      i64 ka, kb;
      gather2<P2>(kmap, k, ka, kb);
      CT a = A[oa + ka];
      CT b = B[ob + kb];
      re = fma(a.x, b.x, fma(-a.y, b.y, re));
      im = fma(a.x, b.y, fma(a.y, b.x, im));
    }
    C[p] = CT{re, im};
  }
}

// Tiled bit-permutation copy for pow2 full-span permutes (the TTGT packs
// and unpack): dst[p] = src[spread(p)]. k_permute_ct reads src scattered
// (one 8/16 B element per lane); here the 6 lowest-src-stride bits are the
// lane index of a coalesced READ and the 6 lowest free dst bits the lane
// index of a coalesced WRITE, staged through a 64x64 XOR-swizzled LDS tile
// (64 KB c128 -> 2 blocks/CU).
#define TN_PERM_AB 6
// BB=5 (64x32 tiles, 32 KB LDS -> more blocks/CU) measured 5.2 TB/s vs
// 4.3 TB/s at 6/6 on the rqc36-interleave shape and holds 5.2 on bit
// reversal (scripts/perm_tune.hip scan, r02)
#define TN_PERM_BB 5
#define TN_PERM_MAXBITS 34

struct PermPerm {
  int rbits;
  u64 restS[TN_PERM_MAXBITS];  // src / dst element strides of each rest bit
  u64 restD[TN_PERM_MAXBITS];
  u64 bS[TN_PERM_BB];  // src strides of the b-bits
  u64 bD[TN_PERM_BB];  // dst strides of the b-bits (ascending)
  u64 aD[TN_PERM_AB];  // dst strides of the a-bits (src strides are 1<<i)
};

template <typename CT>
__global__ __launch_bounds__(512) void k_permute_tile(
    const CT* __restrict__ src, CT* __restrict__ dst, PermPerm pp) {
  constexpr int AN = 1 << TN_PERM_AB, BN = 1 << TN_PERM_BB;
  __shared__ CT tile[AN * BN];
  __shared__ u64 sboffS[BN], sboffD[BN], saoffD[AN];
  const int tid = threadIdx.x;
  if (tid < BN) {
    u64 os = 0, od = 0;
    for (int i = 0; i < TN_PERM_BB; ++i)
      if (tid >> i & 1) {
        os += pp.bS[i];
        od += pp.bD[i];
      }
    sboffS[tid] = os;
    sboffD[tid] = od;
  } else if (tid < BN + AN) {
    const int a = tid - BN;
    u64 od = 0;
    for (int i = 0; i < TN_PERM_AB; ++i)
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
  for (int e = tid; e < AN * BN; e += 512) {
    // lane ~ a: coalesced src reads
    const int a = e & (AN - 1), b = e >> TN_PERM_AB;
    tile[b * AN + ((a ^ b) & (AN - 1))] = src[baseS + sboffS[b] + (u64)a];
  }
  __syncthreads();
  for (int e = tid; e < AN * BN; e += 512) {
    // lane ~ b: coalesced dst writes
    const int b = e & (BN - 1), a = e >> TN_PERM_BB;
    dst[baseD + saoffD[a] + sboffD[b]] = tile[b * AN + ((a ^ b) & (AN - 1))];
  }
}

// fast path: both operands contiguous and identically ordered over the
// contracted legs (the final amplitude dot of two same-legs tensors) —
// pure streaming, no index gather.
template <typename CT>
__global__ void k_dot_partial_linear(const CT* __restrict__ A,
                                     const CT* __restrict__ B,
                                     double2* __restrict__ ws, u64 K) {
  __shared__ double sre[256], sim[256];
  double re = 0.0, im = 0.0;
  for (u64 k = blockIdx.x * (u64)blockDim.x + threadIdx.x; k < K;
       k += gridDim.x * (u64)blockDim.x) {
    CT a = A[k];
    CT b = B[k];
    re = fma((double)a.x, (double)b.x, fma(-(double)a.y, (double)b.y, re));
    im = fma((double)a.x, (double)b.y, fma((double)a.y, (double)b.x, im));
  }
  sre[threadIdx.x] = re;
  sim[threadIdx.x] = im;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < (unsigned)s) {
      sre[threadIdx.x] += sre[threadIdx.x + s];
      sim[threadIdx.x] += sim[threadIdx.x + s];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) ws[blockIdx.x] = make_double2(sre[0], sim[0]);
}

// Tiled dot for a contracted index that is a pure bit-permutation between
// two contiguous pow2 operands (the final amplitude dot of rqc networks:
// both operands hold the same 2^30-ish legs in scrambled orders). A naive
// gather reads one side as random 16B accesses (~1.6 TB/s); here both
// operands are read coalesced and the scramble happens in LDS.
//   a-bits: the 6 lowest A bits (lane index, coalesced A reads)
//   b-bits: the 6 lowest B bits not among the a-bits (coalesced B reads)
//   rest  : the remaining bits (one block per combination)
// Each block stages its 64x64 B tile in LDS (64 KB -> 2 blocks/CU, so one
// block's loads overlap another's compute; the 128x64 variant measured
// 10.7 ms vs 6.2 ms for this at K=2^30 c128; XOR-swizzled so both the
// b-major writes and a-major reads are bank-conflict-free), then streams A.
#define TN_DOT_TILE_ABITS 6
#define TN_DOT_TILE_BBITS 6
#define TN_DOT_TILE_BITS (TN_DOT_TILE_ABITS + TN_DOT_TILE_BBITS)
#define TN_DOT_MAXBITS 34

struct DotPerm {
  int rbits;
  u64 restA[TN_DOT_MAXBITS];  // element strides of each rest bit
  u64 restB[TN_DOT_MAXBITS];
  u64 bA[TN_DOT_TILE_BBITS];  // A strides of the b-bits
  u64 bB[TN_DOT_TILE_BBITS];  // B strides of the b-bits (ascending)
  u64 aB[TN_DOT_TILE_ABITS];  // B strides of the a-bits (A strides are 1<<i)
};

template <typename CT, int BB>
__global__ __launch_bounds__(512) void k_dot_tile(const CT* __restrict__ A,
                                                  const CT* __restrict__ B,
                                                  double2* __restrict__ ws,
                                                  DotPerm dp) {
  constexpr int NB = 1 << BB;
  __shared__ CT tile[NB * 64];
  __shared__ u64 sboffA[NB], sboffB[NB], saoffB[64];
  __shared__ double sred[16];
  const int tid = threadIdx.x;
  if (tid < NB) {
    u64 oa = 0, ob = 0;
    for (int i = 0; i < BB; ++i)
      if (tid >> i & 1) {
        oa += dp.bA[i];
        ob += dp.bB[i];
      }
    sboffA[tid] = oa;
    sboffB[tid] = ob;
  } else if (tid < NB + 64) {
    const int a = tid - NB;
    u64 ob = 0;
    for (int i = 0; i < TN_DOT_TILE_ABITS; ++i)
      if (a >> i & 1) ob += dp.aB[i];
    saoffB[a] = ob;
  }
  u64 baseA = 0, baseB = 0;
  {
    unsigned r = blockIdx.x;
    for (int i = 0; i < dp.rbits; ++i) {
      if (r & 1) {
        baseA += dp.restA[i];
        baseB += dp.restB[i];
      }
      r >>= 1;
    }
  }
  __syncthreads();
  for (int e = tid; e < NB * 64; e += 512) {
    const int b = e & (NB - 1), a = e >> BB;
    tile[b * 64 + (a ^ (b & 63))] = B[baseB + saoffB[a] + sboffB[b]];
  }
  __syncthreads();
  const int wave = tid >> 6, lane = tid & 63;
  double re = 0.0, im = 0.0;
  for (int b = wave; b < NB; b += 8) {
    const CT av = A[baseA + sboffA[b] + (u64)lane];
    const CT bv = tile[b * 64 + (lane ^ (b & 63))];
    re = fma((double)av.x, (double)bv.x, fma(-(double)av.y, (double)bv.y, re));
    im = fma((double)av.x, (double)bv.y, fma((double)av.y, (double)bv.x, im));
  }
  for (int s = 32; s > 0; s >>= 1) {
    re += __shfl_xor(re, s, 64);
    im += __shfl_xor(im, s, 64);
  }
  if (lane == 0) {
    sred[wave * 2] = re;
    sred[wave * 2 + 1] = im;
  }
  __syncthreads();
  if (tid == 0) {
    double tre = 0.0, tim = 0.0;
    for (int w = 0; w < 8; ++w) {
      tre += sred[w * 2];
      tim += sred[w * 2 + 1];
    }
    ws[blockIdx.x] = make_double2(tre, tim);
  }
}

template <bool P2, typename CT>
__global__ void k_dot_partial(const CT* __restrict__ A,
                              const CT* __restrict__ B,
                              double2* __restrict__ ws, u64 K, GatherMap kmap) {
  __shared__ double sre[256], sim[256];
  double re = 0.0, im = 0.0;
  for (u64 k = blockIdx.x * (u64)blockDim.x + threadIdx.x; k < K;
       k += gridDim.x * (u64)blockDim.x) {
    i64 ka, kb;
    gather2<P2>(kmap, k, ka, kb);
    CT a = A[ka];
    CT b = B[kb];
    re = fma((double)a.x, (double)b.x, fma(-(double)a.y, (double)b.y, re));
    im = fma((double)a.x, (double)b.y, fma((double)a.y, (double)b.x, im));
  }
  sre[threadIdx.x] = re;
  sim[threadIdx.x] = im;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < (unsigned)s) {
      sre[threadIdx.x] += sre[threadIdx.x + s];
      sim[threadIdx.x] += sim[threadIdx.x + s];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) ws[blockIdx.x] = make_double2(sre[0], sim[0]);
}

template <typename CT>
__global__ void k_dot_finish(const double2* __restrict__ ws, CT* out,
                             int nblocks) {
  using RT = decltype(CT{}.x);
  __shared__ double sre[256], sim[256];
  double re = 0.0, im = 0.0;
  for (int i = threadIdx.x; i < nblocks; i += blockDim.x) {
    re += ws[i].x;
    im += ws[i].y;
  }
  sre[threadIdx.x] = re;
  sim[threadIdx.x] = im;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < (unsigned)s) {
      sre[threadIdx.x] += sre[threadIdx.x + s];
      sim[threadIdx.x] += sim[threadIdx.x + s];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) *out = CT{(RT)sre[0], (RT)sim[0]};
}

// dst[p] = src[gather(p)] — pack / unpack / general permute (also serves the
// final-tensor Permutor, circuit_builder.rs:89-106, on device).
template <bool P2, typename CT>
__global__ void k_permute_ct(const CT* __restrict__ src,
                             CT* __restrict__ dst, u64 n, GatherMap map) {
  for (u64 p = blockIdx.x * (u64)blockDim.x + threadIdx.x; p < n;
       p += gridDim.x * (u64)blockDim.x) {
    dst[p] = src[gather1<P2>(map, p)];
  }
}

// --- complex GEMM, C[M,N] = A[M,K] @ B[K,N], interleaved c128, row-major ---
// 64x64 C tile per 256-thread block; linearized grid (tile = blockIdx.x,
// row tile = tile / col_tiles) so huge M or N never overflow grid dims.

#define GT 64
#define GK 16

template <typename CT>
__global__ __launch_bounds__(256) void k_zgemm_v1(
    const CT* __restrict__ A, const CT* __restrict__ B,
    CT* __restrict__ C, u64 M, u64 N, u64 K, unsigned col_tiles,
    unsigned tiles, u64 kchunk) {
  __shared__ CT As[GT][GK + 1];
  __shared__ CT Bs[GK][GT + 1];
  const int tx = threadIdx.x % 16, ty = threadIdx.x / 16;
  // 32-bit tile decode (64-bit div emulation costs ~20 VGPRs)
  const unsigned tile = (unsigned)blockIdx.x % tiles;
  const unsigned slice = (unsigned)blockIdx.x / tiles;
  const u64 brow = (u64)(tile / col_tiles) * GT;
  const u64 bcol = (u64)(tile % col_tiles) * GT;
  const u64 kbeg = (u64)slice * kchunk;
  const u64 kend = (kbeg + kchunk < K) ? kbeg + kchunk : K;
  C += (u64)slice * M * N;  // slice 0 == C itself when kchunk == K
  CT acc[4][4];
  for (int i = 0; i < 4; ++i)
    for (int j = 0; j < 4; ++j) acc[i][j] = CT{0, 0};
  for (u64 k0 = kbeg; k0 < kend; k0 += GK) {
    for (int i = threadIdx.x; i < GT * GK; i += 256) {
      int r = i / GK, c = i % GK;
      As[r][c] = (brow + r < M && k0 + c < kend) ? A[(brow + r) * K + k0 + c]
                                                 : CT{0, 0};
    }
    for (int i = threadIdx.x; i < GK * GT; i += 256) {
      int r = i / GT, c = i % GT;
      Bs[r][c] = (k0 + r < kend && bcol + c < N) ? B[(k0 + r) * N + bcol + c]
                                                 : CT{0, 0};
    }
    __syncthreads();
    for (int kk = 0; kk < GK; ++kk) {
      CT a[4], b[4];
      for (int i = 0; i < 4; ++i) a[i] = As[ty * 4 + i][kk];
      for (int j = 0; j < 4; ++j) b[j] = Bs[kk][tx * 4 + j];
      for (int i = 0; i < 4; ++i)
        for (int j = 0; j < 4; ++j) {
          acc[i][j].x = fma(a[i].x, b[j].x, fma(-a[i].y, b[j].y, acc[i][j].x));
          acc[i][j].y = fma(a[i].x, b[j].y, fma(a[i].y, b[j].x, acc[i][j].y));
        }
    }
    __syncthreads();
  }
  for (int i = 0; i < 4; ++i) {
    u64 r = brow + ty * 4 + i;
    if (r >= M) continue;
    for (int j = 0; j < 4; ++j) {
      u64 c = bcol + tx * 4 + j;
      if (c < N) C[r * N + c] = acc[i][j];
    }
  }
}

// MFMA f64 kernel: v_mfma_f64_16x16x4_f64, 8 waves per block, 128x64 tile
// (tile scan on the dominant rqc36 shape: 128x64/8-wave 58.7 TF/s vs
// 64x64/4-wave 56.2; deeper K-tiles and 128-wide tiles lose to occupancy;
// Gauss 3-mult loses to AGPR pressure — scripts/zgemm_tune.hip).
// Wave w owns C rows [w*16, w*16+16); its row slab is 4 column fragments of
// 16x16, each a {re, im} accumulator pair (4 f64 regs each). Per k-quad:
// 4 MFMAs per fragment (Cr += ArBr; Cr += (-Ai)Bi; Ci += ArBi; Ci += AiBr).
// LDS staging is planar with padded rows to avoid bank conflicts.
typedef double v4d __attribute__((ext_vector_type(4)));
typedef float v4f __attribute__((ext_vector_type(4)));

// Per-precision MFMA core. D-fragment row maps verified on hardware
// (scripts/mfma_repro.hip): f64 16x16x4 -> row = 4*reg + lane/16,
// f32 16x16x4 -> row = (lane/16)*4 + reg (they differ!).
template <typename RT>
struct MfmaCore;
template <>
struct MfmaCore<double> {
  using acc_t = v4d;
  static __device__ __forceinline__ acc_t mfma(double a, double b, acc_t c) {
    return __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
  }
  static __device__ __forceinline__ int crow(int lane, int r) {
    return 4 * r + lane / 16;
  }
};
template <>
struct MfmaCore<float> {
  using acc_t = v4f;
  static __device__ __forceinline__ acc_t mfma(float a, float b, acc_t c) {
    return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
  }
  static __device__ __forceinline__ int crow(int lane, int r) {
    return (lane / 16) * 4 + r;
  }
};

#define MF_T 128      // tile rows = MF_WAVES * 16
#define MF_TN 64      // tile cols
#define MF_K 16
#define MF_THREADS 512
#define A_LD 17  // padded row stride (elements) for the A tiles
#define B_LD 66  // padded row stride for the B tiles

template <typename CT>
__global__ __launch_bounds__(MF_THREADS) void k_zgemm_mfma(
    const CT* __restrict__ A, const CT* __restrict__ B,
    CT* __restrict__ C, u64 M, u64 N, u64 K, unsigned col_tiles,
    unsigned tiles, u64 kchunk) {
  using RT = decltype(CT{}.x);
  using Core = MfmaCore<RT>;
  using acc_t = typename Core::acc_t;
  __shared__ RT Ar[MF_T * A_LD];
  __shared__ RT Ai[MF_T * A_LD];
  __shared__ RT Br[MF_K * B_LD];
  __shared__ RT Bi[MF_K * B_LD];

  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const unsigned tile = (unsigned)blockIdx.x % tiles;
  const unsigned slice = (unsigned)blockIdx.x / tiles;
  const u64 brow = (u64)(tile / col_tiles) * MF_T;
  const u64 bcol = (u64)(tile % col_tiles) * MF_TN;
  const u64 kbeg = (u64)slice * kchunk;
  const u64 kend = (kbeg + kchunk < K) ? kbeg + kchunk : K;
  C += (u64)slice * M * N;
  const bool interior = (brow + MF_T <= M) && (bcol + MF_TN <= N);

  acc_t cr[4], ci[4];
  for (int f = 0; f < 4; ++f) {
    cr[f] = acc_t{0, 0, 0, 0};
    ci[f] = acc_t{0, 0, 0, 0};
  }

  // operand map (both precisions): lane l holds A[i = l%16][k = l/16] and
  // B[k = l/16][j = l%16]; the D row map is per-precision (MfmaCore::crow).
  const int fi = lane % 16;
  const int fk = lane / 16;

  for (u64 k0 = kbeg; k0 < kend; k0 += MF_K) {
    for (int i = threadIdx.x; i < MF_T * MF_K; i += MF_THREADS) {
      int r = i / MF_K, c = i % MF_K;
      CT v = (brow + r < M && k0 + c < kend) ? A[(brow + r) * K + k0 + c]
                                             : CT{0, 0};
      Ar[r * A_LD + c] = v.x;
      Ai[r * A_LD + c] = v.y;
    }
    for (int i = threadIdx.x; i < MF_K * MF_TN; i += MF_THREADS) {
      int r = i / MF_TN, c = i % MF_TN;
      CT v = (k0 + r < kend && bcol + c < N) ? B[(k0 + r) * N + bcol + c]
                                             : CT{0, 0};
      Br[r * B_LD + c] = v.x;
      Bi[r * B_LD + c] = v.y;
    }
    __syncthreads();
    for (int kq = 0; kq < MF_K / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + fk;
      const RT ar = Ar[arow * A_LD + ak];
      const RT ai = Ai[arow * A_LD + ak];
      for (int f = 0; f < 4; ++f) {
        const int bcolf = f * 16 + fi;
        const RT br = Br[ak * B_LD + bcolf];
        const RT bi = Bi[ak * B_LD + bcolf];
        cr[f] = Core::mfma(ar, br, cr[f]);
        cr[f] = Core::mfma(-ai, bi, cr[f]);
        ci[f] = Core::mfma(ar, bi, ci[f]);
        ci[f] = Core::mfma(ai, br, ci[f]);
      }
    }
    __syncthreads();
  }

  const int ccol = lane % 16;
  for (int f = 0; f < 4; ++f) {
    for (int r = 0; r < 4; ++r) {
      u64 row = brow + wave * 16 + Core::crow(lane, r);
      u64 col = bcol + f * 16 + ccol;
      if (interior || (row < M && col < N))
        C[row * N + col] = CT{cr[f][r], ci[f][r]};
    }
  }
}


// c128 glds kernel: same 128x64 tile, but LDS-DMA staging into an XOR-
// swizzled interleaved image read back as single ds_read_b128 per operand
// element (re+im together). +20% over the planar-staged kernel on the
// dominant rqc36 shape (70.2-70.7 TF/s = 90% of the 78.6 TF/s f64 spec
// peak; scripts/zgemm_tune.hip — double-buffering loses to occupancy).
// Edge tiles and K-tails use a bounds-guarded staging loop writing the
// identical image (glds cannot mask out-of-range lanes).
__global__ __launch_bounds__(MF_THREADS) void k_zgemm_c128_glds(
    const double2* __restrict__ A, const double2* __restrict__ B,
    double2* __restrict__ C, u64 M, u64 N, u64 K, unsigned col_tiles,
    unsigned tiles, u64 kchunk) {
  constexpr int TM = MF_T, TN = MF_TN, KT = MF_K;
  constexpr int ASLOTS = TM * KT;
  __shared__ double2 As[ASLOTS];  // [r][c ^ (r & 15)]
  __shared__ double2 Bs[KT * TN]; // [k][j ^ ((k & 3) << 4)]
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const unsigned tile = (unsigned)blockIdx.x % tiles;
  const unsigned slice = (unsigned)blockIdx.x / tiles;
  const u64 brow = (u64)(tile / col_tiles) * TM;
  const u64 bcol = (u64)(tile % col_tiles) * TN;
  const u64 kbeg = (u64)slice * kchunk;
  const u64 kend = (kbeg + kchunk < K) ? kbeg + kchunk : K;
  C += (u64)slice * M * N;
  const bool interior = (brow + TM <= M) && (bcol + TN <= N);

  v4d cr[4], ci[4];
  for (int f = 0; f < 4; ++f) {
    cr[f] = v4d{0, 0, 0, 0};
    ci[f] = v4d{0, 0, 0, 0};
  }
  const int fi = lane % 16;

  for (u64 k0 = kbeg; k0 < kend; k0 += KT) {
    if (interior && k0 + KT <= kend) {
      // glds: LDS dst = wave-uniform base + lane*16; per-lane SOURCE
      // pre-swizzled so the image lands linearly.
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
    } else {
      for (int i = threadIdx.x; i < TM * KT; i += MF_THREADS) {
        int r = i / KT, c = i % KT;
        double2 v = (brow + r < M && k0 + c < kend)
                        ? A[(brow + r) * K + k0 + c]
                        : make_double2(0.0, 0.0);
        As[r * KT + (c ^ (r & 15))] = v;
      }
      for (int i = threadIdx.x; i < KT * TN; i += MF_THREADS) {
        int k = i / TN, col = i % TN;
        double2 v = (k0 + k < kend && bcol + col < N)
                        ? B[(k0 + k) * N + bcol + col]
                        : make_double2(0.0, 0.0);
        Bs[k * TN + (col ^ ((k & 3) << 4))] = v;
      }
    }
    __syncthreads();  // carries vmcnt(0): drains the LDS-DMAs
    for (int kq = 0; kq < KT / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + (lane / 16);
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
  for (int f = 0; f < 4; ++f) {
    for (int r = 0; r < 4; ++r) {
      u64 row = brow + crow0 + 4 * r;
      u64 col = bcol + f * 16 + ccol;
      if (interior || (row < M && col < N))
        C[row * N + col] = make_double2(cr[f][r], ci[f][r]);
    }
  }
}


// Pure-glds variant: no guarded branch in the loop (the in-loop fallback
// costs ~20% via register pressure). Launched only when M % 128 == 0,
// N % 64 == 0 and every K-slice is a whole number of 16-tiles.
__global__ __launch_bounds__(MF_THREADS) void k_zgemm_c128_glds_pure(
    const double2* __restrict__ A, const double2* __restrict__ B,
    double2* __restrict__ C, u64 M, u64 N, u64 K, unsigned col_tiles,
    unsigned tiles, u64 kchunk) {
  constexpr int TM = MF_T, TN = MF_TN, KT = MF_K;
  constexpr int ASLOTS = TM * KT;
  __shared__ double2 As[ASLOTS];
  __shared__ double2 Bs[KT * TN];
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  // 32-bit tile decode: 64-bit div emulation costs ~22 VGPRs and a whole
  // wave of occupancy (148 -> 126 VGPRs measured)
  const unsigned tile = (unsigned)blockIdx.x % tiles;
  const unsigned slice = (unsigned)blockIdx.x / tiles;
  const u64 brow = (u64)(tile / col_tiles) * TM;
  const u64 bcol = (u64)(tile % col_tiles) * TN;
  const u64 kbeg = (u64)slice * kchunk;
  const u64 kend = (kbeg + kchunk < K) ? kbeg + kchunk : K;
  C += (u64)slice * M * N;
  v4d cr[4], ci[4];
  for (int f = 0; f < 4; ++f) {
    cr[f] = v4d{0, 0, 0, 0};
    ci[f] = v4d{0, 0, 0, 0};
  }
  const int fi = lane % 16;
  for (u64 k0 = kbeg; k0 < kend; k0 += KT) {
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
    __syncthreads();
    for (int kq = 0; kq < KT / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + (lane / 16);
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
      C[row * N + col] = make_double2(cr[f][r], ci[f][r]);
    }
}


// Gather-staged c128 glds GEMM: reads A and B straight through the TTGT
// pack PERMUTATION instead of from pre-packed copies — the pack kernels
// (and their read+write HBM traffic, ~54 GB per rqc36 contraction)
// disappear. Works because with pow2 dims the packed offset is SEPARABLE:
// src = rowOff(m) + khi(k0) + klo(c), where rowOff/klo are tiny per-block
// LDS tables built in the prologue and khi is a wave-uniform (SALU) sum
// per k-iteration; the per-lane staging address is then two adds, no
// heavier than the multiply it replaces. Everything else (XOR-swizzled
// interleaved LDS image, MFMA body, epilogue) is identical to
// k_zgemm_c128_glds_pure. Launched only for pure shapes (M%128 == 0,
// N%64 == 0, K%16 == 0) with all-pow2 dims.
struct GatherGemmMap {
  int rbits, kbits;   // bits of the row index (M for A / N for B) and of k
  u64 rstride[34];    // source stride (elements) contributed by row bit b
  u64 kstride[34];    // source stride contributed by k bit b
};

__global__ __launch_bounds__(MF_THREADS) void k_zgemm_c128_glds_gather(
    const double2* __restrict__ A, const double2* __restrict__ B,
    double2* __restrict__ C, u64 M, u64 N, u64 K, unsigned col_tiles,
    unsigned tiles, u64 kchunk, GatherGemmMap mA, GatherGemmMap mB) {
  constexpr int TM = MF_T, TN = MF_TN, KT = MF_K;
  __shared__ double2 As[TM * KT];
  __shared__ double2 Bs[KT * TN];
  __shared__ u64 rowOffA[TM];
  __shared__ u64 colOffB[TN];
  __shared__ u64 kloA[KT];
  __shared__ u64 kloB[KT];
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const unsigned tile = (unsigned)blockIdx.x % tiles;
  const unsigned slice = (unsigned)blockIdx.x / tiles;
  const u64 brow = (u64)(tile / col_tiles) * TM;
  const u64 bcol = (u64)(tile % col_tiles) * TN;
  const u64 kbeg = (u64)slice * kchunk;
  const u64 kend = (kbeg + kchunk < K) ? kbeg + kchunk : K;
  C += (u64)slice * M * N;
  // offset tables (once per block)
  for (int t = threadIdx.x; t < TM + TN + 2 * KT; t += MF_THREADS) {
    if (t < TM) {
      u64 idx = brow + t, off = 0;
      for (int b = 0; b < mA.rbits; ++b)
        if ((idx >> b) & 1) off += mA.rstride[b];
      rowOffA[t] = off;
    } else if (t < TM + TN) {
      u64 idx = bcol + (t - TM), off = 0;
      for (int b = 0; b < mB.rbits; ++b)
        if ((idx >> b) & 1) off += mB.rstride[b];
      colOffB[t - TM] = off;
    } else if (t < TM + TN + KT) {
      int c = t - TM - TN;
      u64 off = 0;
      for (int b = 0; b < 4; ++b)
        if ((c >> b) & 1) off += mA.kstride[b];
      kloA[c] = off;
    } else {
      int c = t - TM - TN - KT;
      u64 off = 0;
      for (int b = 0; b < 4; ++b)
        if ((c >> b) & 1) off += mB.kstride[b];
      kloB[c] = off;
    }
  }
  __syncthreads();
  v4d cr[4], ci[4];
  for (int f = 0; f < 4; ++f) {
    cr[f] = v4d{0, 0, 0, 0};
    ci[f] = v4d{0, 0, 0, 0};
  }
  const int fi = lane % 16;
  for (u64 k0 = kbeg; k0 < kend; k0 += KT) {
    u64 khiA = 0, khiB = 0;
    for (int b = 4; b < mA.kbits; ++b)
      if ((k0 >> b) & 1) khiA += mA.kstride[b];
    for (int b = 4; b < mB.kbits; ++b)
      if ((k0 >> b) & 1) khiB += mB.kstride[b];
    const double2* Ak = A + khiA;
    const double2* Bk = B + khiB;
    for (int piece = 0; piece < 4; ++piece) {
      int base = (wave * 4 + piece) * 64;
      int i = base + lane;
      int r = i / KT, c_sw = i % KT;
      int c = c_sw ^ (r & 15);
      const double2* src = &Ak[rowOffA[r] + kloA[c]];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)&As[base], 16, 0, 0);
    }
    for (int piece = 0; piece < 2; ++piece) {
      int base = piece * 512 + wave * 64;
      int j = base + lane;
      int k = j / TN, col_sw = j % TN;
      int col = col_sw ^ ((k & 3) << 4);
      const double2* src = &Bk[kloB[k] + colOffB[col]];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)&Bs[base], 16, 0, 0);
    }
    __syncthreads();
    for (int kq = 0; kq < KT / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + (lane / 16);
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
      C[row * N + col] = make_double2(cr[f][r], ci[f][r]);
    }
}


// c64 pure-glds kernel: one 16-byte LDS-DMA moves TWO float2 elements, so
// the swizzle works at pair granularity (8 pairs per 16-deep K row).
__global__ __launch_bounds__(MF_THREADS) void k_zgemm_c64_glds_pure(
    const float2* __restrict__ A, const float2* __restrict__ B,
    float2* __restrict__ C, u64 M, u64 N, u64 K, unsigned col_tiles,
    unsigned tiles, u64 kchunk) {
  constexpr int TM = MF_T, TN = MF_TN, KT = MF_K;
  __shared__ float2 As[TM * KT];  // [r][2*(cp ^ (r & 7)) + e]
  __shared__ float2 Bs[KT * TN];  // [k][2*(jp ^ ((k & 3) << 2)) + e]
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const unsigned tile = (unsigned)blockIdx.x % tiles;
  const unsigned slice = (unsigned)blockIdx.x / tiles;
  const u64 brow = (u64)(tile / col_tiles) * TM;
  const u64 bcol = (u64)(tile % col_tiles) * TN;
  const u64 kbeg = (u64)slice * kchunk;
  const u64 kend = (kbeg + kchunk < K) ? kbeg + kchunk : K;
  C += (u64)slice * M * N;
  v4f cr[4], ci[4];
  for (int f = 0; f < 4; ++f) {
    cr[f] = v4f{0, 0, 0, 0};
    ci[f] = v4f{0, 0, 0, 0};
  }
  const int fi = lane % 16;
  for (u64 k0 = kbeg; k0 < kend; k0 += KT) {
    // A: 128x16 c64 = 1024 pair-slots; 8 waves x 2 pieces x 64 lanes
    for (int piece = 0; piece < 2; ++piece) {
      int base = (wave * 2 + piece) * 64;  // pair-slot base
      int i = base + lane;
      int r = i / (KT / 2), cp_sw = i % (KT / 2);
      int cp = cp_sw ^ (r & 7);
      const float2* src = &A[(brow + r) * K + k0 + 2 * cp];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)&As[2 * base], 16, 0, 0);
    }
    // B: 16x64 c64 = 512 pair-slots; 8 waves x 1 piece
    {
      int base = wave * 64;
      int j = base + lane;
      int k = j / (TN / 2), jp_sw = j % (TN / 2);
      int jp = jp_sw ^ ((k & 3) << 2);
      const float2* src = &B[(k0 + k) * N + bcol + 2 * jp];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)&Bs[2 * base], 16, 0, 0);
    }
    __syncthreads();
    for (int kq = 0; kq < KT / 4; ++kq) {
      const int arow = wave * 16 + fi;
      const int ak = kq * 4 + (lane / 16);
      float2 a = As[arow * KT + 2 * ((ak / 2) ^ (arow & 7)) + (ak & 1)];
      for (int f = 0; f < 4; ++f) {
        const int bcolf = f * 16 + fi;
        float2 b = Bs[ak * TN + 2 * ((bcolf / 2) ^ ((ak & 3) << 2)) +
                      (bcolf & 1)];
        cr[f] = __builtin_amdgcn_mfma_f32_16x16x4f32(a.x, b.x, cr[f], 0, 0, 0);
        cr[f] = __builtin_amdgcn_mfma_f32_16x16x4f32(-a.y, b.y, cr[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f32_16x16x4f32(a.x, b.y, ci[f], 0, 0, 0);
        ci[f] = __builtin_amdgcn_mfma_f32_16x16x4f32(a.y, b.x, ci[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  const int crow0 = wave * 16 + (lane / 16) * 4;  // f32 D map: (l/16)*4+r
  const int ccol = lane % 16;
  for (int f = 0; f < 4; ++f)
    for (int r = 0; r < 4; ++r) {
      u64 row = brow + crow0 + r;
      u64 col = bcol + f * 16 + ccol;
      C[row * N + col] = make_float2(cr[f][r], ci[f][r]);
    }
}

// split-K reduce: C[p] = sum over slices of ws[s][p]
template <typename CT>
__global__ void k_splitk_reduce(const CT* __restrict__ ws,
                                CT* __restrict__ C, u64 nout, int slices) {
  using RT = decltype(CT{}.x);
  for (u64 p = blockIdx.x * (u64)blockDim.x + threadIdx.x; p < nout;
       p += gridDim.x * (u64)blockDim.x) {
    RT re = 0, im = 0;
    for (int s = 0; s < slices; ++s) {
      CT v = ws[(u64)s * nout + p];
      re += v.x;
      im += v.y;
    }
    C[p] = CT{re, im};
  }
}

// ---------------------------------------------------------------------------
// host-side planning
// ---------------------------------------------------------------------------

struct Meta {
  int nd;
  u64 labels[TN_MAXR];
  u64 dims[TN_MAXR];
  i64 strides[TN_MAXR];  // in elements
  const void* data;
};

struct AxisInfo {
  u64 dim;
  i64 sa;
  i64 sb;
};

static int log2_u64(u64 v) {
  int s = 0;
  while ((1ull << s) < v) ++s;
  return s;
}

static bool axes_pow2(const std::vector<AxisInfo>& axes) {
  for (const auto& a : axes)
    if (a.dim & (a.dim - 1)) return false;
  return true;
}

// Encode a map. The pow2 (shift/mask) and general (div/mod) ENCODINGS are
// incompatible; when a kernel takes several maps the caller must pick ONE
// encoding for all of them (allow_pow2 = the joint decision).
static int build_map(const std::vector<AxisInfo>& axes, GatherMap* m,
                     bool allow_pow2 = true) {
  int n = (int)axes.size();
  if (n > TN_MAXR) return -1;
  m->n = n;
  u64 pstride = 1;
  bool p2 = allow_pow2 && axes_pow2(axes);
  for (int i = n - 1; i >= 0; --i) {
    m->pstride[i] = pstride;
    m->dim[i] = axes[i].dim;
    m->sa[i] = axes[i].sa;
    m->sb[i] = axes[i].sb;
    pstride *= axes[i].dim;
  }
  m->pow2 = p2 ? 1 : 0;
  if (p2) {
    for (int i = 0; i < n; ++i) {
      m->pstride[i] = (u64)log2_u64(m->pstride[i]);
      m->dim[i] = m->dim[i] - 1;
    }
  }
  return 0;
}

static void ensure_mempool(int device) {
  static bool done[64] = {};
  if (device >= 0 && device < 64 && !done[device]) {
    hipMemPool_t pool;
    if (hipDeviceGetDefaultMemPool(&pool, device) == hipSuccess) {
      uint64_t threshold = UINT64_MAX;
      (void)hipMemPoolSetAttribute(pool, hipMemPoolAttrReleaseThreshold,
                                   &threshold);
    }
    done[device] = true;
  }
}

struct StepStats {
  int kind;  // 0 smallk/anyk, 1 dot, 2 gemm, 3 gemm+unpack
  u64 m, n, k;
  hipEvent_t gemm_ev0 = nullptr;  // when set, record around the GEMM launch
  hipEvent_t gemm_ev1 = nullptr;
};

// --- device arena: one reserved slab, host-side first-fit bookkeeping ---
// Reuse is stream-ordered (all users launch on one stream), so a freed
// block can be handed out immediately: the consuming kernels are ordered
// behind the producing ones. Avoids hipMallocAsync's per-allocation page
// mapping, which costs seconds per contraction at 30-70 GB intermediates.
struct Arena {
  char* base = nullptr;
  size_t size = 0;
  struct Block {
    size_t off, sz;
    bool free;
  };
  std::vector<Block> blocks;

  int reserve(size_t bytes) {
    if (base) return 0;
    // never try to reserve more than the device can hold: clamp to 90% of
    // free memory so per-step spill allocations still have headroom
    size_t free_b = 0, total_b = 0;
    if (hipMemGetInfo(&free_b, &total_b) == hipSuccess && free_b > 0) {
      size_t cap = free_b - free_b / 10;
      if (bytes > cap) bytes = cap;
    }
    if (hipMalloc((void**)&base, bytes) != hipSuccess) {
      (void)hipGetLastError();  // swallow the sticky OOM; caller recovers
      return -1;
    }
    size = bytes;
    blocks = {{0, bytes, true}};
    return 0;
  }
  void* alloc(size_t bytes) {
    if (!base) return nullptr;
    bytes = (bytes + 255) & ~(size_t)255;
    for (size_t i = 0; i < blocks.size(); ++i) {
      if (blocks[i].free && blocks[i].sz >= bytes) {
        size_t off = blocks[i].off;
        if (blocks[i].sz > bytes) {
          blocks.insert(blocks.begin() + i + 1,
                        {off + bytes, blocks[i].sz - bytes, true});
        }
        blocks[i].sz = bytes;
        blocks[i].free = false;
        return base + off;
      }
    }
    return nullptr;  // exhausted -> caller falls back
  }
  bool owns(const void* p) const {
    return base && p >= base && p < base + size;
  }
  void release(void* p) {
    size_t off = (char*)p - base;
    for (size_t i = 0; i < blocks.size(); ++i) {
      if (blocks[i].off == off) {
        blocks[i].free = true;
        // merge with next, then previous
        if (i + 1 < blocks.size() && blocks[i + 1].free) {
          blocks[i].sz += blocks[i + 1].sz;
          blocks.erase(blocks.begin() + i + 1);
        }
        if (i > 0 && blocks[i - 1].free) {
          blocks[i - 1].sz += blocks[i].sz;
          blocks.erase(blocks.begin() + i);
        }
        return;
      }
    }
  }
  void destroy() {
    if (base) (void)hipFree(base);
    base = nullptr;
    blocks.clear();
  }
};

// workspace allocation context threaded through the einsum implementation
struct WsCtx {
  Arena* arena = nullptr;  // optional
  hipStream_t stream = nullptr;
  // stream-capture mode (hipGraph): non-arena allocations poison the graph
  // (their pointers would dangle on replay) and frees must be deferred past
  // EndCapture (hipFree device-syncs, which is illegal mid-capture)
  bool capturing = false;
  bool spilled = false;
  std::vector<void*>* deferred = nullptr;
  // pack-pipeline stream (null = pipelining off): pack permutes of the
  // TTGT route may run K-window-chunked on this stream, overlapped with
  // the window GEMMs on `stream`
  hipStream_t stream2 = nullptr;
  // deferred event destruction (capture mode); null = destroy immediately
  std::vector<hipEvent_t>* events = nullptr;
};

// Non-arena workspace uses plain hipMalloc, NOT hipMallocAsync: on this
// ROCm stack the stream-ordered pool intermittently loses kernel writes to
// freshly-expanded pool pages (trailing rows of a GEMM output read back as
// zeros even with AMD_SERIALIZE_KERNEL=3; plain hipMalloc is always clean).
// The hot path (tn_net) uses the Arena slab, so the sync cost lands only on
// the standalone einsum entry points and the arena-exhaustion fallback.
static int ws_alloc(WsCtx& ctx, void** p, size_t bytes) {
  if (ctx.arena) {
    *p = ctx.arena->alloc(bytes);
    if (*p) return TN_OK;
  }
  if (ctx.capturing) ctx.spilled = true;
  if (hipMalloc(p, bytes) != hipSuccess) {
    (void)hipGetLastError();  // don't leave a sticky OOM for launch checks
    size_t fb = 0, tb = 0;
    (void)hipMemGetInfo(&fb, &tb);
    char buf[192];
    snprintf(buf, sizeof buf,
             "device allocation failed (%zu bytes; %zu free of %zu; arena %s)",
             bytes, fb, tb, ctx.arena ? "exhausted" : "absent");
    g_last_error = buf;
    return TN_ERR_OOM;
  }
  return TN_OK;
}

static void ws_free(WsCtx& ctx, void* p) {
  if (!p) return;
  if (ctx.arena && ctx.arena->owns(p)) {
    ctx.arena->release(p);
    return;
  }
  if (ctx.capturing && ctx.deferred) {
    ctx.deferred->push_back(p);
    return;
  }
  (void)hipFree(p);  // hipFree device-syncs before releasing the pages
}

static int grid_for(u64 nout, int block = 256) {
  u64 blocks = (nout + (u64)block - 1) / block;
  u64 cap = 64 * 2048;  // grid-stride covers the remainder
  if (blocks > cap) blocks = cap;
  if (blocks == 0) blocks = 1;
  return (int)blocks;
}

// Launch the tiled permute when the axis list describes a pow2 full-span
// bit permutation (contiguous source, >= 2^20 elements); returns false to
// fall back to the gather permute (strided views, non-pow2, tiny sizes).
template <typename CT>
static bool permute_tiled(const CT* src, CT* dst, u64 elems,
                          const std::vector<AxisInfo>& ax,
                          hipStream_t stream) {
  if (elems < (1ull << 20) || elems > (1ull << TN_PERM_MAXBITS)) return false;
  struct BitSD {
    u64 ss, sd;
  };
  std::vector<BitSD> bits;
  u64 dstride = 1;
  for (int i = (int)ax.size() - 1; i >= 0; --i) {
    const AxisInfo& a = ax[i];
    if (a.dim == 1) continue;  // contributes no bits; stride irrelevant
    if ((a.dim & (a.dim - 1)) || a.sa <= 0) return false;
    for (u64 d = 1; d < a.dim; d <<= 1)
      bits.push_back({(u64)a.sa * d, dstride * d});
    dstride *= a.dim;
  }
  const int nb = (int)bits.size();
  if (nb <= TN_PERM_AB + TN_PERM_BB || (1ull << nb) != elems) return false;
  std::vector<int> byS(nb), byD(nb);
  for (int i = 0; i < nb; ++i) byS[i] = byD[i] = i;
  std::sort(byS.begin(), byS.end(),
            [&](int x, int y) { return bits[x].ss < bits[y].ss; });
  std::sort(byD.begin(), byD.end(),
            [&](int x, int y) { return bits[x].sd < bits[y].sd; });
  // the a-bits (lane index) must be the TN_PERM_AB lowest SOURCE strides,
  // exactly 1..2^(AB-1), for coalesced reads. All other bits may carry
  // ARBITRARY source strides (the kernel sums per-bit offsets), which
  // admits sub-box permutes — e.g. the pack pipeline's K-windows, whose
  // fixed leading legs leave gaps in the source span. The destination
  // side is a full span by construction (suffix products above).
  for (int i = 0; i < TN_PERM_AB; ++i)
    if (bits[byS[i]].ss != (1ull << i)) return false;
  PermPerm pp;
  std::vector<char> used(nb, 0);
  for (int i = 0; i < TN_PERM_AB; ++i) {
    used[byS[i]] = 1;
    pp.aD[i] = bits[byS[i]].sd;
  }
  int nbb = 0, nr = 0;
  for (int i = 0; i < nb && nbb < TN_PERM_BB; ++i) {
    const int bi = byD[i];
    if (used[bi]) continue;
    pp.bS[nbb] = bits[bi].ss;
    pp.bD[nbb] = bits[bi].sd;
    used[bi] = 1;
    ++nbb;
  }
  for (int i = 0; i < nb; ++i) {
    if (used[byS[i]]) continue;
    pp.restS[nr] = bits[byS[i]].ss;
    pp.restD[nr] = bits[byS[i]].sd;
    ++nr;
  }
  pp.rbits = nr;
  k_permute_tile<<<dim3(1u << nr), 512, 0, stream>>>(src, dst, pp);
  return true;
}

// Build the bit-scatter map of a packed [row][k] (or [k][col]) layout:
// packed-index bit b -> source stride. Axes are outer..inner; the inner
// axis supplies the low bits. Fails on non-pow2 dims or non-positive
// strides (the gather GEMM then falls back to the pack path).
static bool build_ggmap(const Meta& t, const std::vector<int>& axes_r,
                        const std::vector<int>& axes_k, GatherGemmMap* m) {
  auto fill = [&](const std::vector<int>& axes, u64* stride, int* nbits) {
    int nb = 0;
    for (int x = (int)axes.size() - 1; x >= 0; --x) {
      u64 d = t.dims[axes[x]];
      if (d & (d - 1)) return false;
      if (d > 1 && t.strides[axes[x]] <= 0) return false;
      for (u64 v = 1; v < d; v <<= 1) {
        if (nb >= 34) return false;
        stride[nb++] = (u64)t.strides[axes[x]] * v;
      }
    }
    *nbits = nb;
    return true;
  };
  return fill(axes_r, m->rstride, &m->rbits) &&
         fill(axes_k, m->kstride, &m->kbits);
}

static bool pipeline_disabled() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("TN_NO_PIPELINE");
    v = (e && e[0] && e[0] != '0') ? 1 : 0;
  }
  return v == 1;
}

// testing hook: run every shape-feasible step through the pipeline,
// ignoring the profitability gate (exercises the window/event machinery
// on small cases the gate would reject)
static bool pipeline_forced() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("TN_PIPELINE_FORCE");
    v = (e && e[0] && e[0] != '0') ? 1 : 0;
  }
  return v == 1;
}

// The gather-staged GEMM is OPT-IN (TN_GATHER_GEMM=1): measured on the
// rqc36 dominant shapes it is ~35% SLOWER than pack + pure GEMM (44.9 vs
// 69.6 TF/s on M16384/N4096/K32768) because the GEMM re-reads its operand
// strips ~12x across tiles — the pack is a bandwidth AMORTIZER (scattered
// source read once, packed copy re-read coalesced), not removable
// overhead. Kept for low-reuse shapes and as a recorded negative result
// (DESIGN.md "Round-2 measured state").
static bool gather_gemm_disabled() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("TN_GATHER_GEMM");
    v = (e && e[0] && e[0] != '0') ? 0 : 1;
  }
  return v == 1;
}

// Launch the best permute kernel for a gather described by `ax` (tiled
// bit-permutation when applicable, else the index-gather permute).
template <typename CT>
static int launch_permute(const CT* src, CT* dst, u64 elems,
                          const std::vector<AxisInfo>& ax,
                          hipStream_t stream) {
  if (!permute_tiled(src, dst, elems, ax, stream)) {
    GatherMap map;
    if (build_map(ax, &map)) FAILV(TN_ERR_INVALID, "rank too large");
    int blocks = grid_for(elems);
    if (map.pow2)
      k_permute_ct<true><<<blocks, 256, 0, stream>>>(src, dst, elems, map);
    else
      k_permute_ct<false><<<blocks, 256, 0, stream>>>(src, dst, elems, map);
  }
  HIP_CHECK(hipGetLastError());
  return TN_OK;
}

// core einsum over device buffers; out is contiguous row-major in out order.
template <typename CT>
static int einsum_dev_impl(const u64* out_labels, const u64* out_shape,
                           int out_nd, const Meta& A, const Meta& B,
                           CT* out, hipStream_t stream, WsCtx& ws,
                           StepStats* stats) {
  const CT* Adata = (const CT*)A.data;
  const CT* Bdata = (const CT*)B.data;
  if (A.nd > TN_MAXR || B.nd > TN_MAXR || out_nd > TN_MAXR)
    FAILV(TN_ERR_INVALID, "tensor rank exceeds %d", TN_MAXR);
  auto find = [](const Meta& t, u64 lab) {
    for (int i = 0; i < t.nd; ++i)
      if (t.labels[i] == lab) return i;
    return -1;
  };
  int apos[TN_MAXR], bpos[TN_MAXR];
  u64 M = 1, N = 1, K = 1;
  std::vector<int> m_out, n_out;  // positions (in out) of M legs / N legs
  for (int i = 0; i < out_nd; ++i) {
    apos[i] = find(A, out_labels[i]);
    bpos[i] = find(B, out_labels[i]);
    if (apos[i] >= 0 && bpos[i] >= 0)
      FAILV(TN_ERR_INVALID, "out label %llu present in both inputs",
            (unsigned long long)out_labels[i]);
    if (apos[i] >= 0) {
      if (A.dims[apos[i]] != out_shape[i])
        FAILV(TN_ERR_INVALID, "dim mismatch on out label %llu",
              (unsigned long long)out_labels[i]);
      M *= out_shape[i];
      m_out.push_back(i);
    } else if (bpos[i] >= 0) {
      if (B.dims[bpos[i]] != out_shape[i])
        FAILV(TN_ERR_INVALID, "dim mismatch on out label %llu",
              (unsigned long long)out_labels[i]);
      N *= out_shape[i];
      n_out.push_back(i);
    } else {
      FAILV(TN_ERR_INVALID, "out label %llu not found in inputs",
            (unsigned long long)out_labels[i]);
    }
  }
  std::vector<int> k_a, k_b;  // K legs, A order
  for (int i = 0; i < A.nd; ++i) {
    int j = find(B, A.labels[i]);
    if (j >= 0) {
      for (int o = 0; o < out_nd; ++o)
        if (out_labels[o] == A.labels[i])
          FAILV(TN_ERR_INVALID, "contracted label %llu appears in out",
                (unsigned long long)A.labels[i]);
      if (A.dims[i] != B.dims[j])
        FAILV(TN_ERR_INVALID, "K dim mismatch on label %llu",
              (unsigned long long)A.labels[i]);
      K *= A.dims[i];
      k_a.push_back(i);
      k_b.push_back(j);
    }
  }
  // every input label must be contracted (shared) or appear in out — traces
  // and repeated labels are outside the tensor_mult boundary contract
  // (contraction.rs:88-116 always passes the symmetric difference)
  auto in_out = [&](u64 lab) {
    for (int o = 0; o < out_nd; ++o)
      if (out_labels[o] == lab) return true;
    return false;
  };
  for (int i = 0; i < A.nd; ++i)
    if (find(B, A.labels[i]) < 0 && !in_out(A.labels[i]))
      FAILV(TN_ERR_INVALID, "label %llu of A neither contracted nor in out",
            (unsigned long long)A.labels[i]);
  for (int i = 0; i < B.nd; ++i)
    if (find(A, B.labels[i]) < 0 && !in_out(B.labels[i]))
      FAILV(TN_ERR_INVALID, "label %llu of B neither contracted nor in out",
            (unsigned long long)B.labels[i]);
  u64 nout = M * N;
  if (stats) {
    stats->m = M;
    stats->n = N;
    stats->k = K;
    stats->kind = 0;
  }

  // ---- dot: scalar output, large K ----
  if (M == 1 && N == 1 && K > TN_SMALLK) {
    std::vector<AxisInfo> kax;
    for (size_t t = 0; t < k_a.size(); ++t)
      kax.push_back({A.dims[k_a[t]], A.strides[k_a[t]], B.strides[k_b[t]]});
    GatherMap kmap;
    if (build_map(kax, &kmap)) FAILV(TN_ERR_INVALID, "rank too large");
    // linear iff each axis' source stride equals its packed suffix product
    bool linear = true;
    {
      i64 pstride = 1;
      for (int i = (int)kax.size() - 1; i >= 0; --i) {
        if (kax[i].sa != pstride || kax[i].sb != pstride) linear = false;
        pstride *= (i64)kax[i].dim;
      }
    }
    // tiled bit-permutation path: both operands are contiguous pow2 spans
    // of the same K elements in different axis orders
    DotPerm dp;
    int tbits = -1;
    if (!linear && K >= (1ull << 20)) {
      struct BitSB {
        u64 sa, sb;
      };
      std::vector<BitSB> bits;
      bool ok = true;
      for (auto& ax : kax) {
        if (ax.dim == 1) continue;  // contributes no bits
        if ((ax.dim & (ax.dim - 1)) || ax.sa <= 0 || ax.sb <= 0) {
          ok = false;
          break;
        }
        for (u64 d = 1; d < ax.dim; d <<= 1)
          bits.push_back({(u64)ax.sa * d, (u64)ax.sb * d});
      }
      const int bb = TN_DOT_TILE_BBITS;
      const int nb = (int)bits.size();
      if (ok && nb <= TN_DOT_MAXBITS && nb > TN_DOT_TILE_ABITS + bb) {
        std::vector<int> byA(nb), byB(nb);
        for (int i = 0; i < nb; ++i) byA[i] = byB[i] = i;
        std::sort(byA.begin(), byA.end(),
                  [&](int x, int y) { return bits[x].sa < bits[y].sa; });
        std::sort(byB.begin(), byB.end(),
                  [&](int x, int y) { return bits[x].sb < bits[y].sb; });
        for (int i = 0; i < nb; ++i)
          if (bits[byA[i]].sa != (1ull << i) || bits[byB[i]].sb != (1ull << i))
            ok = false;  // not a contiguous span on both sides
        if (ok) {
          std::vector<char> used(nb, 0);
          for (int i = 0; i < TN_DOT_TILE_ABITS; ++i) {
            used[byA[i]] = 1;
            dp.aB[i] = bits[byA[i]].sb;
          }
          int nbb = 0, nr = 0;
          for (int i = 0; i < nb && nbb < bb; ++i) {
            const int bi = byB[i];
            if (used[bi]) continue;
            dp.bA[nbb] = bits[bi].sa;
            dp.bB[nbb] = bits[bi].sb;
            used[bi] = 1;
            ++nbb;
          }
          for (int i = 0; i < nb; ++i) {
            if (used[byA[i]]) continue;
            dp.restA[nr] = bits[byA[i]].sa;
            dp.restB[nr] = bits[byA[i]].sb;
            ++nr;
          }
          dp.rbits = nr;
          tbits = nr;
        }
      }
    }
    if (tbits > 0) {
      if (stats) stats->kind = 6;  // tiled bit-permutation dot
      const u64 nblk = 1ull << tbits;
      double2* wsbuf;
      {
        int rc_ = ws_alloc(ws, (void**)&wsbuf, nblk * sizeof(double2));
        if (rc_) return rc_;
      }
      k_dot_tile<CT, TN_DOT_TILE_BBITS><<<dim3((unsigned)nblk), 512, 0,
                                           stream>>>(Adata, Bdata, wsbuf, dp);
      k_dot_finish<<<1, 256, 0, stream>>>(wsbuf, out, (int)nblk);
      ws_free(ws, wsbuf);
      HIP_CHECK(hipGetLastError());
      return TN_OK;
    }
    if (stats) stats->kind = linear ? 5 : 1;  // 5 = linear (streaming) dot
    int blocks = grid_for(K);
    if (blocks > 2048) blocks = 2048;
    double2* wsbuf;
    {
      int rc_ = ws_alloc(ws, (void**)&wsbuf, blocks * sizeof(double2));
      if (rc_) return rc_;
    }
    if (linear)
      k_dot_partial_linear<<<blocks, 256, 0, stream>>>(Adata, Bdata, wsbuf, K);
    else if (kmap.pow2)
      k_dot_partial<true><<<blocks, 256, 0, stream>>>(Adata, Bdata, wsbuf, K,
                                                      kmap);
    else
      k_dot_partial<false><<<blocks, 256, 0, stream>>>(Adata, Bdata, wsbuf, K,
                                                       kmap);
    k_dot_finish<<<1, 256, 0, stream>>>(wsbuf, out, blocks);
    ws_free(ws, wsbuf);
    HIP_CHECK(hipGetLastError());
    return TN_OK;
  }

  // ---- smallk / anyk: small K, or skinny GEMM shapes ----
  // mid-K big shapes (e.g. M=2^20 N=2^10 K=32) run ~3x faster as a packed
  // MFMA GEMM than as an address-gather stream
  bool skinny = (M < 16 || N < 16);
  bool gemm_worthy = (K >= 16 && M >= MF_T && N >= MF_TN);
  bool gather_ok = (K <= TN_SMALLK || skinny);
  auto run_gather = [&]() -> int {
    if (stats) stats->kind = 0;
    // out map: every out axis, with its source stride in A or B
    std::vector<AxisInfo> oax;
    for (int i = 0; i < out_nd; ++i) {
      AxisInfo ax;
      ax.dim = out_shape[i];
      ax.sa = apos[i] >= 0 ? A.strides[apos[i]] : 0;
      ax.sb = bpos[i] >= 0 ? B.strides[bpos[i]] : 0;
      oax.push_back(ax);
    }
    std::vector<AxisInfo> kax;
    for (size_t t = 0; t < k_a.size(); ++t)
      kax.push_back({A.dims[k_a[t]], A.strides[k_a[t]], B.strides[k_b[t]]});
    if (kax.empty()) kax.push_back({1, 0, 0});
    bool p2 = axes_pow2(oax) && axes_pow2(kax);
    GatherMap omap, kmap;
    if (build_map(oax, &omap, p2) || build_map(kax, &kmap, p2))
      FAILV(TN_ERR_INVALID, "rank too large");
    int blocks = grid_for(nout);
    if (K <= TN_SMALLK) {
      // table decode pays once ~4 offsets/thread are amortized over >=8
      // elements and the per-element decode is actually deep
      if (p2 && nout >= (1ull << 26) && nout < (1ull << 32) && omap.n >= 8)
        k_einsum_smallk_tbl<<<blocks > 32768 ? 32768 : blocks, 256, 0,
                              stream>>>(Adata, Bdata, out, nout, omap, kmap,
                                        (int)K);
      else if (p2)
        k_einsum_smallk<true><<<blocks, 256, 0, stream>>>(
            Adata, Bdata, out, nout, omap, kmap, (int)K);
      else
        k_einsum_smallk<false><<<blocks, 256, 0, stream>>>(
            Adata, Bdata, out, nout, omap, kmap, (int)K);
    } else {
      if (p2)
        k_einsum_anyk<true><<<blocks, 256, 0, stream>>>(Adata, Bdata, out,
                                                        nout, omap, kmap, K);
      else
        k_einsum_anyk<false><<<blocks, 256, 0, stream>>>(Adata, Bdata, out,
                                                         nout, omap, kmap, K);
    }
    HIP_CHECK(hipGetLastError());
    return TN_OK;
  };
  if (gather_ok && !gemm_worthy) return run_gather();

  // ---- TTGT: pack (if needed) + GEMM + unpack (if needed) ----
  if (stats) stats->kind = 2;
  // GEMM operand layouts: A' = [M legs in out order][K legs in A order],
  // B' = [K legs in A order][N legs in out order]. C = [M..][N..]; equal to
  // `out` iff the M legs all precede the N legs there (always true for the
  // executor's symmetric-difference order).
  std::vector<int> a_axes;  // A axis order for A'
  for (int p : m_out) a_axes.push_back(apos[p]);
  for (int i : k_a) a_axes.push_back(i);
  std::vector<int> b_axes;  // B axis order for B'
  for (int j : k_b) b_axes.push_back(j);
  for (int p : n_out) b_axes.push_back(bpos[p]);

  auto is_ready = [](const Meta& t, const std::vector<int>& axes) {
    if ((int)axes.size() != t.nd) return false;
    i64 stride = 1;
    for (int i = (int)axes.size() - 1; i >= 0; --i) {
      if (t.strides[axes[i]] != stride) return false;
      stride *= (i64)t.dims[axes[i]];
    }
    return true;
  };

  const bool needA = !is_ready(A, a_axes);
  const bool needB = !is_ready(B, b_axes);
  const bool mfma_shape = (M >= 32 && N >= 32);
  // gather-staged GEMM (c128): fold the pack permutes into the GEMM's LDS
  // staging when the shape is pure and every dim is pow2 — no pack
  // kernels, no pack workspace, no pack HBM traffic
  GatherGemmMap gmA{}, gmB{};
  bool gather_gemm = false;
  if constexpr (std::is_same_v<CT, double2>) {
    if ((needA || needB) && mfma_shape && (M % MF_T == 0) &&
        (N % MF_TN == 0) && (K % MF_K == 0) && !gather_gemm_disabled()) {
      std::vector<int> a_m, b_n;
      for (int p : m_out) a_m.push_back(apos[p]);
      for (int p : n_out) b_n.push_back(bpos[p]);
      gather_gemm = build_ggmap(A, a_m, k_a, &gmA) &&
                    build_ggmap(B, b_n, k_b, &gmB);
    }
  }

  // does C == out directly? (decided before packing so the pack-pipeline
  // below can target the right C buffer)
  bool direct = true;
  if (!m_out.empty() && !n_out.empty() && m_out.back() > n_out.front())
    direct = false;
  CT* Cg = out;
  CT* tmpC = nullptr;
  if (!direct) {
    if (stats) stats->kind = 3;
    {
      int rc_ = ws_alloc(ws, (void**)&tmpC, nout * sizeof(CT));
      if (rc_ == TN_ERR_OOM && gather_ok) return run_gather();
      if (rc_) return rc_;
    }
    Cg = tmpC;
  }

  // ---- pack-pipeline: chunk the pack over K windows and overlap each
  // window's permute (stream2) with the previous window's GEMM (stream).
  // The window GEMMs write split-K-style slices reduced at the end, so no
  // kernel changes are needed; the pack permutes (HBM-bound, ~4 TB/s)
  // ride in the MFMA-bound GEMM's spare bandwidth. Applied when the model
  // predicts a net win over the serial pack (the slice buffers cost
  // (2p)·M·N extra traffic). Requires a packed A (a ready A's K-windows
  // are strided); B may be packed or ready (row windows are contiguous).
  bool pipelined = false;
  if (ws.stream2 && needA && mfma_shape && (M % MF_T == 0) &&
      (N % MF_TN == 0) && (K % MF_K == 0) && !pipeline_disabled()) {
    u64 packbytes =
        ((needA ? M * K : 0) + (needB ? K * N : 0)) * sizeof(CT);
    double best_save = 0.0;
    int best_p = 0, best_npre = 0, feas_p = 0, feas_npre = 0;
    u64 best_kc = 0, feas_kc = 0;
    {
      u64 p = 1;
      u64 tiles_w = (M / MF_T) * ((N + MF_TN - 1) / MF_TN);
      for (size_t x = 0; x < k_a.size() && p < 16; ++x) {
        p *= A.dims[k_a[x]];
        if (p < 2 || p > 16) continue;
        u64 kc = K / p;
        if (kc % MF_K || tiles_w < 256) continue;
        double save = 2.0 * (double)packbytes / 4e12 * (1.0 - 1.0 / p) -
                      2.0 * p * (double)(M * N * sizeof(CT)) / 6e12;
        feas_p = (int)p;
        feas_kc = kc;
        feas_npre = (int)(x + 1);
        if (save > best_save) {
          best_save = save;
          best_p = (int)p;
          best_kc = kc;
          best_npre = (int)(x + 1);
        }
      }
    }
    // measured gate (r02 A/B on hardware): marginal steps LOSE to the
    // serial pack (window-launch overhead + slice-reduce traffic), so
    // pipeline only when the predicted win is substantial and the pack
    // dwarfs the output (rqc36's 4-10x-ratio steps regressed ~12 ms;
    // syc49's 64x step gains ~12 ms)
    if (pipeline_forced() && feas_p) {
      best_p = feas_p;
      best_kc = feas_kc;
      best_npre = feas_npre;
    } else if (best_p &&
               (best_save <= 5e-3 ||
                (double)packbytes < 8.0 * (double)(M * N * sizeof(CT))))
      best_p = 0;
    if (best_p) {
      const int P = best_p;
      const u64 kc = best_kc;
      CT* Awin = nullptr;
      CT* Bwin = nullptr;
      CT* slices = nullptr;
      int rc_ = ws_alloc(ws, (void**)&Awin, M * K * sizeof(CT));
      if (rc_ == TN_OK && needB)
        rc_ = ws_alloc(ws, (void**)&Bwin, K * N * sizeof(CT));
      if (rc_ == TN_OK)
        rc_ = ws_alloc(ws, (void**)&slices, (u64)P * nout * sizeof(CT));
      hipEvent_t e0 = nullptr, ep[16] = {};
      bool ev_ok = (rc_ == TN_OK);
      if (ev_ok && hipEventCreate(&e0) != hipSuccess) {
        e0 = nullptr;
        ev_ok = false;
      }
      for (int w = 0; ev_ok && w < P; ++w)
        if (hipEventCreate(&ep[w]) != hipSuccess) {
          ep[w] = nullptr;
          ev_ok = false;
        }
      if (!ev_ok) {
        // shortage: fall through to the serial pack path
        ws_free(ws, Awin);
        ws_free(ws, Bwin);
        ws_free(ws, slices);
        if (ws.events) {
          if (e0) ws.events->push_back(e0);
          for (int w = 0; w < P; ++w)
            if (ep[w]) ws.events->push_back(ep[w]);
        } else {
          if (e0) (void)hipEventDestroy(e0);
          for (int w = 0; w < P; ++w)
            if (ep[w]) (void)hipEventDestroy(ep[w]);
        }
      } else {
        // mixed-radix suffix products of the leading K legs (window digit
        // x has radix A.dims[k_a[x]])
        u64 sufA[16] = {};
        {
          u64 s = 1;
          for (int x = best_npre - 1; x >= 0; --x) {
            sufA[x] = s;
            s *= A.dims[k_a[x]];
          }
        }
        HIP_CHECK(hipEventRecord(e0, stream));
        HIP_CHECK(hipStreamWaitEvent(ws.stream2, e0, 0));
        for (int w = 0; w < P; ++w) {
          // source offsets of window w
          u64 offA = 0, offB = 0;
          for (int x = 0; x < best_npre; ++x) {
            u64 digit = ((u64)w / sufA[x]) % A.dims[k_a[x]];
            offA += digit * (u64)A.strides[k_a[x]];
            offB += digit * (u64)B.strides[k_b[x]];
          }
          {
            std::vector<AxisInfo> ax;
            for (int p2 : m_out)
              ax.push_back({A.dims[apos[p2]], A.strides[apos[p2]], 0});
            for (size_t x = best_npre; x < k_a.size(); ++x)
              ax.push_back({A.dims[k_a[x]], A.strides[k_a[x]], 0});
            int prc = launch_permute(Adata + offA, Awin + (u64)w * M * kc,
                                     M * kc, ax, ws.stream2);
            if (prc) return prc;
          }
          if (needB) {
            std::vector<AxisInfo> ax;
            for (size_t x = best_npre; x < k_b.size(); ++x)
              ax.push_back({B.dims[k_b[x]], B.strides[k_b[x]], 0});
            for (int p2 : n_out)
              ax.push_back({B.dims[bpos[p2]], B.strides[bpos[p2]], 0});
            int prc = launch_permute(Bdata + offB, Bwin + (u64)w * kc * N,
                                     kc * N, ax, ws.stream2);
            if (prc) return prc;
          }
          HIP_CHECK(hipEventRecord(ep[w], ws.stream2));
        }
        u64 tiles_w = (M / MF_T) * (N / MF_TN);
        if (stats && stats->gemm_ev0)
          HIP_CHECK(hipEventRecord(stats->gemm_ev0, stream));
        for (int w = 0; w < P; ++w) {
          HIP_CHECK(hipStreamWaitEvent(stream, ep[w], 0));
          const CT* Aw = Awin + (u64)w * M * kc;
          const CT* Bw = needB ? Bwin + (u64)w * kc * N
                               : Bdata + (u64)w * kc * N;
          CT* Cw = slices + (u64)w * nout;
          if constexpr (std::is_same_v<CT, double2>) {
            k_zgemm_c128_glds_pure<<<dim3((unsigned)tiles_w), MF_THREADS, 0,
                                     stream>>>(
                Aw, Bw, Cw, M, N, kc, (unsigned)(N / MF_TN),
                (unsigned)tiles_w, kc);
          } else {
            k_zgemm_c64_glds_pure<<<dim3((unsigned)tiles_w), MF_THREADS, 0,
                                    stream>>>(
                (const float2*)Aw, (const float2*)Bw, (float2*)Cw, M, N, kc,
                (unsigned)(N / MF_TN), (unsigned)tiles_w, kc);
          }
        }
        k_splitk_reduce<<<grid_for(nout), 256, 0, stream>>>(slices, Cg,
                                                            nout, P);
        if (stats && stats->gemm_ev1)
          HIP_CHECK(hipEventRecord(stats->gemm_ev1, stream));
        HIP_CHECK(hipGetLastError());
        ws_free(ws, Awin);
        ws_free(ws, Bwin);
        ws_free(ws, slices);
        if (ws.events) {
          ws.events->push_back(e0);
          for (int w = 0; w < P; ++w) ws.events->push_back(ep[w]);
        } else {
          (void)hipEventDestroy(e0);
          for (int w = 0; w < P; ++w) (void)hipEventDestroy(ep[w]);
        }
        pipelined = true;
      }
    }
  }
  if (pipelined) {
    if (!direct) {
      std::vector<i64> tmp_stride(out_nd, 0);
      i64 stride = 1;
      for (int t = (int)n_out.size() - 1; t >= 0; --t) {
        tmp_stride[n_out[t]] = stride;
        stride *= (i64)out_shape[n_out[t]];
      }
      for (int t = (int)m_out.size() - 1; t >= 0; --t) {
        tmp_stride[m_out[t]] = stride;
        stride *= (i64)out_shape[m_out[t]];
      }
      std::vector<AxisInfo> ax;
      for (int i = 0; i < out_nd; ++i)
        ax.push_back({out_shape[i], tmp_stride[i], 0});
      int rc_ = launch_permute((const CT*)tmpC, out, nout, ax, stream);
      if (rc_) return rc_;
    }
    ws_free(ws, tmpC);
    return TN_OK;
  }

  const CT* Ag = Adata;
  const CT* Bg = Bdata;
  CT* packA = nullptr;
  CT* packB = nullptr;
  if (!gather_gemm && needA) {
    std::vector<AxisInfo> ax;
    for (int axis : a_axes) ax.push_back({A.dims[axis], A.strides[axis], 0});
    u64 elems = M * K;
    {
      int rc_ = ws_alloc(ws, (void**)&packA, elems * sizeof(CT));
      // pack workspace doesn't fit -> run the shape through the (slower)
      // gather kernels instead of failing the whole contraction
      if (rc_ == TN_ERR_OOM && gather_ok) {
        ws_free(ws, tmpC);
        return run_gather();
      }
      if (rc_) return rc_;
    }
    {
      int rc_ = launch_permute(Adata, packA, elems, ax, stream);
      if (rc_) return rc_;
    }
    Ag = packA;
  }
  if (!gather_gemm && needB) {
    std::vector<AxisInfo> ax;
    for (int axis : b_axes) ax.push_back({B.dims[axis], B.strides[axis], 0});
    u64 elems = K * N;
    {
      int rc_ = ws_alloc(ws, (void**)&packB, elems * sizeof(CT));
      if (rc_ == TN_ERR_OOM && gather_ok) {
        ws_free(ws, packA);
        ws_free(ws, tmpC);
        return run_gather();
      }
      if (rc_) return rc_;
    }
    {
      int rc_ = launch_permute(Bdata, packB, elems, ax, stream);
      if (rc_) return rc_;
    }
    Bg = packB;
  }

  bool mfma = mfma_shape;
  u64 row_tile_h = mfma ? MF_T : GT;
  u64 row_tiles = (M + row_tile_h - 1) / row_tile_h;
  u64 col_tiles = (N + GT - 1) / GT;
  u64 tiles = row_tiles * col_tiles;
  // split-K: with few tiles and deep K, slice K across extra blocks into
  // partial buffers + a reduce pass (fills the 256 CUs; a 64-tile K=2^20
  // GEMM goes from ~6 to ~50 TFLOP/s)
  u64 splitk = 1;
  if (tiles < 192 && K >= 256) {
    while (tiles * splitk < 256 && K / (splitk * 2) >= 64) splitk *= 2;
    if (splitk > 64) splitk = 64;
  }
  u64 kchunk = (K + splitk - 1) / splitk;
  kchunk = ((kchunk + MF_K - 1) / MF_K) * MF_K;  // tile-aligned
  splitk = (K + kchunk - 1) / kchunk;
  CT* gemm_out = Cg;
  CT* splitbuf = nullptr;
  if (splitk > 1) {
    int rc_ = ws_alloc(ws, (void**)&splitbuf, splitk * nout * sizeof(CT));
    if (rc_ == TN_ERR_OOM) {
      splitk = 1;  // split-K is an optimization; run unsplit when tight
      kchunk = ((K + MF_K - 1) / MF_K) * MF_K;
    } else if (rc_) {
      return rc_;
    } else {
      gemm_out = splitbuf;
    }
  }
  dim3 grid((unsigned)(tiles * splitk));
  if (stats && stats->gemm_ev0)
    HIP_CHECK(hipEventRecord(stats->gemm_ev0, stream));
  if (mfma) {
    if constexpr (std::is_same_v<CT, double2>) {
      bool pure = (M % MF_T == 0) && (N % MF_TN == 0) && (K % MF_K == 0) &&
                  (kchunk % MF_K == 0);
      // gather_gemm implies pure: its precondition covers M/N/K tiling and
      // kchunk is MF_K-rounded above
      if (gather_gemm)
        k_zgemm_c128_glds_gather<<<grid, MF_THREADS, 0, stream>>>(
            Adata, Bdata, gemm_out, M, N, K, (unsigned)col_tiles,
            (unsigned)tiles, kchunk, gmA, gmB);
      else if (pure)
        k_zgemm_c128_glds_pure<<<grid, MF_THREADS, 0, stream>>>(
            Ag, Bg, gemm_out, M, N, K, (unsigned)col_tiles, (unsigned)tiles,
            kchunk);
      else
        k_zgemm_c128_glds<<<grid, MF_THREADS, 0, stream>>>(
            Ag, Bg, gemm_out, M, N, K, (unsigned)col_tiles, (unsigned)tiles,
            kchunk);
    } else {
      bool pure = (M % MF_T == 0) && (N % MF_TN == 0) && (K % MF_K == 0) &&
                  (kchunk % MF_K == 0);
      if (pure)
        k_zgemm_c64_glds_pure<<<grid, MF_THREADS, 0, stream>>>(
            (const float2*)Ag, (const float2*)Bg, (float2*)gemm_out, M, N, K,
            (unsigned)col_tiles, (unsigned)tiles, kchunk);
      else
        k_zgemm_mfma<<<grid, MF_THREADS, 0, stream>>>(
            Ag, Bg, gemm_out, M, N, K, (unsigned)col_tiles, (unsigned)tiles,
            kchunk);
    }
  } else {
    k_zgemm_v1<<<grid, 256, 0, stream>>>(Ag, Bg, gemm_out, M, N, K,
                                         (unsigned)col_tiles, (unsigned)tiles,
                                         kchunk);
  }
  if (splitk > 1)
    k_splitk_reduce<<<grid_for(nout), 256, 0, stream>>>(splitbuf, Cg, nout,
                                                        (int)splitk);
  if (stats && stats->gemm_ev1)
    HIP_CHECK(hipEventRecord(stats->gemm_ev1, stream));
  HIP_CHECK(hipGetLastError());
  if (splitbuf) ws_free(ws, splitbuf);

  if (!direct) {
    // permute tmp [M legs (out order)][N legs (out order)] -> out order
    // source strides of each out axis inside tmp:
    std::vector<i64> tmp_stride(out_nd, 0);
    i64 stride = 1;
    for (int t = (int)n_out.size() - 1; t >= 0; --t) {
      tmp_stride[n_out[t]] = stride;
      stride *= (i64)out_shape[n_out[t]];
    }
    for (int t = (int)m_out.size() - 1; t >= 0; --t) {
      tmp_stride[m_out[t]] = stride;
      stride *= (i64)out_shape[m_out[t]];
    }
    std::vector<AxisInfo> ax;
    for (int i = 0; i < out_nd; ++i)
      ax.push_back({out_shape[i], tmp_stride[i], 0});
    {
      int rc_ = launch_permute((const CT*)tmpC, out, nout, ax, stream);
      if (rc_) return rc_;
    }
  }
  ws_free(ws, packA);
  ws_free(ws, packB);
  ws_free(ws, tmpC);
  return TN_OK;
}

// ---------------------------------------------------------------------------
// C API: device management + einsum entry points
// ---------------------------------------------------------------------------

static int g_device = 0;

extern "C" int tn_device_count(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

extern "C" int tn_set_device(int device) {
  HIP_CHECK(hipSetDevice(device));
  g_device = device;
  ensure_mempool(device);
  return TN_OK;
}

static int require_gpu() {
  if (tn_device_count() == 0)
    FAILV(TN_ERR_NO_GPU,
          "no AMD GPU present — tnc_hip has no CPU fallback by design");
  return TN_OK;
}

static int fill_meta(Meta* m, const u64* labels, const u64* shape,
                     const i64* strides, const void* data, size_t nd) {
  if (nd > TN_MAXR) FAILV(TN_ERR_INVALID, "rank %zu exceeds %d", nd, TN_MAXR);
  m->nd = (int)nd;
  i64 contig = 1;
  for (int i = (int)nd - 1; i >= 0; --i) {
    m->labels[i] = labels[i];
    m->dims[i] = shape[i];
    m->strides[i] = strides ? strides[i] : contig;
    contig *= (i64)shape[i];
  }
  m->data = (const double2*)data;
  return TN_OK;
}

template <typename CT>
static int einsum_dev_entry(const u64* out_labels, const u64* out_shape,
                            size_t out_ndim, const u64* a_labels,
                            const u64* a_shape, const i64* a_strides,
                            const void* a_dev, size_t a_ndim,
                            const u64* b_labels, const u64* b_shape,
                            const i64* b_strides, const void* b_dev,
                            size_t b_ndim, void* out_dev, void* stream) {
  int rc = require_gpu();
  if (rc) return rc;
  Meta A, B;
  rc = fill_meta(&A, a_labels, a_shape, a_strides, a_dev, a_ndim);
  if (rc) return rc;
  rc = fill_meta(&B, b_labels, b_shape, b_strides, b_dev, b_ndim);
  if (rc) return rc;
  WsCtx ws{nullptr, (hipStream_t)stream};
  return einsum_dev_impl<CT>(out_labels, out_shape, (int)out_ndim, A, B,
                             (CT*)out_dev, (hipStream_t)stream, ws, nullptr);
}

extern "C" int tn_einsum_c128_dev(const u64* out_labels, const u64* out_shape,
                                  size_t out_ndim, const u64* a_labels,
                                  const u64* a_shape, const i64* a_strides,
                                  const void* a_dev, size_t a_ndim,
                                  const u64* b_labels, const u64* b_shape,
                                  const i64* b_strides, const void* b_dev,
                                  size_t b_ndim, void* out_dev, void* stream) {
  return einsum_dev_entry<double2>(out_labels, out_shape, out_ndim, a_labels,
                                   a_shape, a_strides, a_dev, a_ndim, b_labels,
                                   b_shape, b_strides, b_dev, b_ndim, out_dev,
                                   stream);
}

extern "C" int tn_einsum_c64_dev(const u64* out_labels, const u64* out_shape,
                                 size_t out_ndim, const u64* a_labels,
                                 const u64* a_shape, const i64* a_strides,
                                 const void* a_dev, size_t a_ndim,
                                 const u64* b_labels, const u64* b_shape,
                                 const i64* b_strides, const void* b_dev,
                                 size_t b_ndim, void* out_dev, void* stream) {
  return einsum_dev_entry<float2>(out_labels, out_shape, out_ndim, a_labels,
                                  a_shape, a_strides, a_dev, a_ndim, b_labels,
                                  b_shape, b_strides, b_dev, b_ndim, out_dev,
                                  stream);
}

static u64 span_elems(const u64* shape, const i64* strides, size_t nd) {
  // max linear offset + 1, assuming nonnegative strides
  u64 span = 1;
  for (size_t i = 0; i < nd; ++i)
    span += (shape[i] - 1) * (u64)(strides ? strides[i] : 0);
  if (!strides) {
    span = 1;
    for (size_t i = 0; i < nd; ++i) span *= shape[i];
  }
  return span;
}

template <typename CT>
static int einsum_host_entry(const u64* out_labels, const u64* out_shape,
                             size_t out_ndim, const u64* a_labels,
                             const u64* a_shape, const i64* a_strides,
                             const void* a_data, size_t a_ndim,
                             const u64* b_labels, const u64* b_shape,
                             const i64* b_strides, const void* b_data,
                             size_t b_ndim, void* out_data) {
  int rc = require_gpu();
  if (rc) return rc;
  HIP_CHECK(hipSetDevice(g_device));
  ensure_mempool(g_device);
  const size_t es = sizeof(CT);
  if (a_strides)
    for (size_t i = 0; i < a_ndim; ++i)
      if (a_strides[i] < 0) FAILV(TN_ERR_INVALID, "negative strides");
  if (b_strides)
    for (size_t i = 0; i < b_ndim; ++i)
      if (b_strides[i] < 0) FAILV(TN_ERR_INVALID, "negative strides");
  u64 a_span = span_elems(a_shape, a_strides, a_ndim);
  u64 b_span = span_elems(b_shape, b_strides, b_ndim);
  u64 out_elems = 1;
  for (size_t i = 0; i < out_ndim; ++i) out_elems *= out_shape[i];
  void *da, *db, *dout;
  HIP_CHECK(hipMalloc(&da, a_span * es));
  HIP_CHECK(hipMalloc(&db, b_span * es));
  HIP_CHECK(hipMalloc(&dout, out_elems * es));
  HIP_CHECK(hipMemcpy(da, a_data, a_span * es, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(db, b_data, b_span * es, hipMemcpyHostToDevice));
  rc = einsum_dev_entry<CT>(out_labels, out_shape, out_ndim, a_labels,
                            a_shape, a_strides, da, a_ndim, b_labels, b_shape,
                            b_strides, db, b_ndim, dout, nullptr);
  if (rc == TN_OK) {
    HIP_CHECK(hipMemcpy(out_data, dout, out_elems * es,
                        hipMemcpyDeviceToHost));
  }
  (void)hipFree(da);
  (void)hipFree(db);
  (void)hipFree(dout);
  return rc;
}

extern "C" int tn_einsum_c128(const u64* out_labels, const u64* out_shape,
                              size_t out_ndim, const u64* a_labels,
                              const u64* a_shape, const i64* a_strides,
                              const void* a_data, size_t a_ndim,
                              const u64* b_labels, const u64* b_shape,
                              const i64* b_strides, const void* b_data,
                              size_t b_ndim, void* out_data) {
  return einsum_host_entry<double2>(out_labels, out_shape, out_ndim, a_labels,
                                    a_shape, a_strides, a_data, a_ndim,
                                    b_labels, b_shape, b_strides, b_data,
                                    b_ndim, out_data);
}

extern "C" int tn_einsum_c64(const u64* out_labels, const u64* out_shape,
                             size_t out_ndim, const u64* a_labels,
                             const u64* a_shape, const i64* a_strides,
                             const void* a_data, size_t a_ndim,
                             const u64* b_labels, const u64* b_shape,
                             const i64* b_strides, const void* b_data,
                             size_t b_ndim, void* out_data) {
  return einsum_host_entry<float2>(out_labels, out_shape, out_ndim, a_labels,
                                   a_shape, a_strides, a_data, a_ndim,
                                   b_labels, b_shape, b_strides, b_data,
                                   b_ndim, out_data);
}

// ---------------------------------------------------------------------------
// network executor (contract_tensor_network replacement)
// ---------------------------------------------------------------------------

struct DevTensor {
  std::vector<u64> labels;
  std::vector<u64> dims;
  void* data = nullptr;
  u64 elems = 1;
  bool owned = false;     // intermediate owned by the walk (freed on consume)
  bool external = false;  // caller-owned device buffer (never freed by us)
};

struct tn_net {
  int device = 0;
  int dtype = 0;   // 0 = c128, 1 = c64
  size_t esize = 16;
  hipStream_t stream = nullptr;
  hipStream_t stream2 = nullptr;  // pack-overlap stream (prepack permutes)
  std::vector<DevTensor> leaves;
  DevTensor final_t;       // final tensor of the last contract (owned)
  bool final_in_arena = false;
  bool has_final = false;
  Arena arena;
  u64 pool_in_use = 0;
  // hipGraph replay of the (launch-bound) walk: captured after the first
  // normal run when every workspace block came from the arena, so the
  // graph's baked device pointers are stable across replays (the arena's
  // host-side first-fit is deterministic and the final block stays
  // reserved). Invalidated by a different path or new leaves.
  hipGraphExec_t graph_exec = nullptr;
  std::vector<u64> graph_pairs;
  int graph_state = 0;  // 0 = no normal run yet, 1 = ready to capture,
                        // 2 = captured, -1 = disabled (spill/failure)

  void invalidate_graph() {
    if (graph_exec) (void)hipGraphExecDestroy(graph_exec);
    graph_exec = nullptr;
    graph_pairs.clear();
    if (graph_state > 0) graph_state = 0;
  }
};

extern "C" tn_net* tn_net_create2(int device, int dtype) {
  if (dtype != 0 && dtype != 1) {
    g_last_error = "dtype must be 0 (c128) or 1 (c64)";
    return nullptr;
  }
  if (tn_device_count() == 0) {
    g_last_error = "no AMD GPU present — tnc_hip has no CPU fallback by design";
    return nullptr;
  }
  if (hipSetDevice(device) != hipSuccess) {
    g_last_error = "hipSetDevice failed";
    return nullptr;
  }
  ensure_mempool(device);
  tn_net* net = new tn_net();
  net->device = device;
  net->dtype = dtype;
  net->esize = dtype == 0 ? 16 : 8;
  if (hipStreamCreate(&net->stream) != hipSuccess) {
    delete net;
    g_last_error = "hipStreamCreate failed";
    return nullptr;
  }
  if (hipStreamCreate(&net->stream2) != hipSuccess) {
    (void)hipStreamDestroy(net->stream);
    delete net;
    g_last_error = "hipStreamCreate failed";
    return nullptr;
  }
  return net;
}

extern "C" tn_net* tn_net_create(int device) {
  return tn_net_create2(device, 0);
}

extern "C" int tn_net_reserve(tn_net* net, uint64_t bytes) {
  if (!net) FAILV(TN_ERR_INVALID, "null net");
  net->invalidate_graph();
  HIP_CHECK(hipSetDevice(net->device));
  if (net->arena.reserve(bytes))
    FAILV(TN_ERR_OOM, "arena reservation of %llu bytes failed",
          (unsigned long long)bytes);
  return TN_OK;
}

extern "C" int64_t tn_net_add_leaf(tn_net* net, const u64* labels,
                                   const u64* dims, size_t ndim,
                                   const void* host_data) {
  if (!net) return -TN_ERR_INVALID;
  if (ndim > TN_MAXR) {
    g_last_error = "rank too large";
    return -TN_ERR_INVALID;
  }
  net->invalidate_graph();  // the network changed
  if (hipSetDevice(net->device) != hipSuccess) return -TN_ERR_HIP;
  DevTensor t;
  t.labels.assign(labels, labels + ndim);
  t.dims.assign(dims, dims + ndim);
  for (size_t i = 0; i < ndim; ++i) t.elems *= dims[i];
  if (hipMalloc((void**)&t.data, t.elems * net->esize) != hipSuccess) {
    g_last_error = "hipMalloc failed for leaf";
    return -TN_ERR_OOM;
  }
  if (hipMemcpy(t.data, host_data, t.elems * net->esize,
                hipMemcpyHostToDevice) != hipSuccess) {
    (void)hipFree(t.data);
    g_last_error = "hipMemcpy failed for leaf";
    return -TN_ERR_HIP;
  }
  t.owned = false;  // leaves persist; freed only at destroy
  net->leaves.push_back(std::move(t));
  return (int64_t)net->leaves.size() - 1;
}

extern "C" int64_t tn_net_add_leaf_dev(tn_net* net, const u64* labels,
                                       const u64* dims, size_t ndim,
                                       void* dev_data) {
  if (!net) return -TN_ERR_INVALID;
  if (ndim > TN_MAXR) {
    g_last_error = "rank too large";
    return -TN_ERR_INVALID;
  }
  net->invalidate_graph();  // the network changed
  DevTensor t;
  t.labels.assign(labels, labels + ndim);
  t.dims.assign(dims, dims + ndim);
  for (size_t i = 0; i < ndim; ++i) t.elems *= dims[i];
  t.data = dev_data;
  t.owned = false;
  t.external = true;
  net->leaves.push_back(std::move(t));
  return (int64_t)net->leaves.size() - 1;
}

// out legs of one step = symmetric difference, A-only legs in A order then
// B-only in B order (tensor.rs:709-725) — the executor's layout contract.
static void symdiff(const DevTensor& a, const DevTensor& b,
                    std::vector<u64>* labels, std::vector<u64>* dims) {
  for (size_t i = 0; i < a.labels.size(); ++i) {
    bool shared = false;
    for (u64 bl : b.labels)
      if (bl == a.labels[i]) shared = true;
    if (!shared) {
      labels->push_back(a.labels[i]);
      dims->push_back(a.dims[i]);
    }
  }
  for (size_t i = 0; i < b.labels.size(); ++i) {
    bool shared = false;
    for (u64 al : a.labels)
      if (al == b.labels[i]) shared = true;
    if (!shared) {
      labels->push_back(b.labels[i]);
      dims->push_back(b.dims[i]);
    }
  }
}

// ---- pack-overlap planning ------------------------------------------------
// Mirror of einsum_dev_impl's dispatch, host-side and launch-free: would the
// step take the TTGT route, and which operands would it pack? Used by the
// walk to pre-permute the NEXT step's GEMM operands on a second stream while
// the current step's GEMM runs (the GEMMs are MFMA-bound and use <10% of the
// HBM bandwidth the permutes need, so the packs ride along ~free). Any
// divergence from the real dispatch is performance-only: a prepacked operand
// is a valid contiguous tensor in pack order, every kernel class consumes it
// correctly, and an operand NOT prepacked is packed inline as before.
struct PackPlan {
  bool packA = false, packB = false;
  std::vector<int> a_axes, b_axes;  // axis orders of the packed layouts
  u64 MK = 0, KN = 0;               // packed element counts
};

static bool plan_prepack(const Meta& A, const Meta& B, const u64* out_labels,
                         const u64* out_shape, int out_nd, PackPlan* plan,
                         bool c128 = false) {
  if (A.nd > TN_MAXR || B.nd > TN_MAXR || out_nd > TN_MAXR) return false;
  auto find = [](const Meta& t, u64 lab) {
    for (int i = 0; i < t.nd; ++i)
      if (t.labels[i] == lab) return i;
    return -1;
  };
  int apos[TN_MAXR], bpos[TN_MAXR];
  u64 M = 1, N = 1, K = 1;
  std::vector<int> m_out, n_out;
  for (int i = 0; i < out_nd; ++i) {
    apos[i] = find(A, out_labels[i]);
    bpos[i] = find(B, out_labels[i]);
    if (apos[i] >= 0 && bpos[i] >= 0) return false;
    if (apos[i] >= 0) {
      M *= out_shape[i];
      m_out.push_back(i);
    } else if (bpos[i] >= 0) {
      N *= out_shape[i];
      n_out.push_back(i);
    } else {
      return false;
    }
  }
  std::vector<int> k_a, k_b;
  for (int i = 0; i < A.nd; ++i) {
    int j = find(B, A.labels[i]);
    if (j >= 0) {
      K *= A.dims[i];
      k_a.push_back(i);
      k_b.push_back(j);
    }
  }
  if (M == 1 && N == 1 && K > TN_SMALLK) return false;  // dot route
  bool skinny = (M < 16 || N < 16);
  bool gemm_worthy = (K >= 16 && M >= MF_T && N >= MF_TN);
  bool gather_ok = (K <= TN_SMALLK || skinny);
  if (gather_ok && !gemm_worthy) return false;  // gather route
  // steps the gather-staged GEMM will take need no pack at all
  if (c128 && (M >= 32 && N >= 32) && (M % MF_T == 0) && (N % MF_TN == 0) &&
      (K % MF_K == 0) && !gather_gemm_disabled()) {
    bool pow2 = true;
    for (int i = 0; i < A.nd; ++i)
      if ((A.dims[i] & (A.dims[i] - 1)) ||
          (A.dims[i] > 1 && A.strides[i] <= 0))
        pow2 = false;
    for (int i = 0; i < B.nd; ++i)
      if ((B.dims[i] & (B.dims[i] - 1)) ||
          (B.dims[i] > 1 && B.strides[i] <= 0))
        pow2 = false;
    if (pow2) return false;
  }
  std::vector<int> a_axes, b_axes;
  for (int p : m_out) a_axes.push_back(apos[p]);
  for (int i : k_a) a_axes.push_back(i);
  for (int j : k_b) b_axes.push_back(j);
  for (int p : n_out) b_axes.push_back(bpos[p]);
  auto is_ready = [](const Meta& t, const std::vector<int>& axes) {
    if ((int)axes.size() != t.nd) return false;
    i64 stride = 1;
    for (int i = (int)axes.size() - 1; i >= 0; --i) {
      if (t.strides[axes[i]] != stride) return false;
      stride *= (i64)t.dims[axes[i]];
    }
    return true;
  };
  plan->packA = !is_ready(A, a_axes);
  plan->packB = !is_ready(B, b_axes);
  plan->a_axes = std::move(a_axes);
  plan->b_axes = std::move(b_axes);
  plan->MK = M * K;
  plan->KN = K * N;
  return plan->packA || plan->packB;
}

// fill a Meta from a contiguous row-major DevTensor
static void meta_of(const DevTensor& t, Meta* m) {
  m->nd = (int)t.labels.size();
  for (int x = 0; x < m->nd; ++x) {
    m->labels[x] = t.labels[x];
    m->dims[x] = t.dims[x];
  }
  i64 stride = 1;
  for (int x = m->nd - 1; x >= 0; --x) {
    m->strides[x] = stride;
    stride *= (i64)m->dims[x];
  }
  m->data = t.data;
}

static bool prepack_disabled() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("TN_NO_PREPACK");
    v = (e && e[0] && e[0] != '0') ? 1 : 0;
  }
  return v == 1;
}

static int contract_impl(tn_net* net, const u64* pairs, size_t nsteps,
                         double* step_ms, double* gemm_ms, int32_t* kind,
                         double* elapsed_ms, bool capture = false,
                         std::vector<void*>* deferred = nullptr,
                         bool* spilled = nullptr) {
  if (!net) FAILV(TN_ERR_INVALID, "null net");
  HIP_CHECK(hipSetDevice(net->device));
  WsCtx ws{net->arena.base ? &net->arena : nullptr, net->stream, capture,
           false, deferred};
  ws.stream2 = net->stream2;
  if (net->has_final && net->final_t.owned) {
    if (net->final_in_arena)
      net->arena.release(net->final_t.data);
    else
      (void)hipFree(net->final_t.data);
    net->final_t = DevTensor();
    net->has_final = false;
    net->final_in_arena = false;
  }
  size_t n = net->leaves.size();
  std::vector<DevTensor> slots(net->leaves);  // shallow copies; owned=false
  for (auto& s : slots) s.owned = false;
  std::vector<char> alive(n, 1);

  bool profiled = (step_ms != nullptr);
  std::vector<hipEvent_t> ev;
  std::vector<int> kinds_local;
  if (profiled) {
    ev.resize(4 * nsteps);  // step start/end + gemm start/end
    for (auto& e : ev) HIP_CHECK(hipEventCreate(&e));
    kinds_local.assign(nsteps, 0);
  }

  double t0_ms = 0.0;
  hipEvent_t walk_start = nullptr, walk_end = nullptr;
  if (!capture) {
    HIP_CHECK(hipStreamSynchronize(net->stream));
    HIP_CHECK(hipEventCreate(&walk_start));
    HIP_CHECK(hipEventCreate(&walk_end));
    HIP_CHECK(hipEventRecord(walk_start, net->stream));
  }

  int rc = TN_OK;
  // pack-overlap state: while step s runs on net->stream, the NEXT step's
  // GEMM pack permutes run on net->stream2 (after an event marking steps
  // < s complete). The replaced originals are freed at the top of step s+1,
  // after the main stream waits on the packs — so every arena block's
  // next user is ordered behind its last reader, preserving the arena's
  // stream-ordered reuse contract across both streams.
  struct Pending {
    void* old_data;
    bool old_owned;
  };
  std::vector<Pending> pending;
  hipEvent_t pending_done = nullptr;
  std::vector<hipEvent_t> prep_events;
  ws.events = &prep_events;  // pipeline events: destroyed after the walk
  const bool overlap =
      !prepack_disabled() && net->stream2 && net->arena.base != nullptr;

  for (size_t s = 0; s < nsteps; ++s) {
    u64 i = pairs[2 * s], j = pairs[2 * s + 1];
    if (i >= n || j >= n || !alive[i] || !alive[j] || i == j) {
      rc = TN_ERR_INVALID;
      g_last_error = "invalid contraction path step";
      break;
    }
    // join this step's prepack (issued during step s-1): the packed
    // operands replace slots[i]/slots[j]; free the originals only now,
    // behind the wait, so their blocks cannot be re-handed while the pack
    // kernels still read them
    if (pending_done) {
      if (hipStreamWaitEvent(net->stream, pending_done, 0) != hipSuccess) {
        rc = TN_ERR_HIP;
        g_last_error = "hipStreamWaitEvent failed (prepack join)";
        break;
      }
      for (auto& p : pending)
        if (p.old_owned && p.old_data) ws_free(ws, p.old_data);
      pending.clear();
      pending_done = nullptr;
    }
    DevTensor& A = slots[i];
    DevTensor& B = slots[j];
    DevTensor out;
    symdiff(A, B, &out.labels, &out.dims);
    out.elems = 1;
    for (u64 d : out.dims) out.elems *= d;
    if (ws_alloc(ws, (void**)&out.data, out.elems * net->esize) != TN_OK) {
      rc = TN_ERR_OOM;
      g_last_error = "device allocation failed for intermediate";
      break;
    }
    out.owned = true;

    // prepack the next step's GEMM operands on stream2 (overlaps this
    // step's kernels). A prepacked slot becomes a contiguous tensor in
    // pack order — einsum_dev_impl then sees is_ready and skips its
    // inline pack. Arena-only; on any shortage the step simply packs
    // inline as before.
    if (overlap && s + 1 < nsteps) {
      u64 ni = pairs[2 * (s + 1)], nj = pairs[2 * (s + 1) + 1];
      bool valid = ni < n && nj < n && ni != nj && ni != j && nj != j &&
                   (ni == i || alive[ni]) && (nj == i || alive[nj]);
      if (valid) {
        const DevTensor& TA = (ni == i) ? out : slots[ni];
        const DevTensor& TB = (nj == i) ? out : slots[nj];
        std::vector<u64> nlabels, ndims;
        symdiff(TA, TB, &nlabels, &ndims);
        Meta mna, mnb;
        PackPlan plan;
        if ((int)TA.labels.size() <= TN_MAXR &&
            (int)TB.labels.size() <= TN_MAXR &&
            (int)nlabels.size() <= TN_MAXR) {
          meta_of(TA, &mna);
          meta_of(TB, &mnb);
          if (plan_prepack(mna, mnb, nlabels.data(), ndims.data(),
                           (int)nlabels.size(), &plan,
                           net->dtype == 0)) {
            void* dstA = nullptr;
            void* dstB = nullptr;
            bool wantA = plan.packA && ni != i;
            bool wantB = plan.packB && nj != i;
            if (wantA) dstA = net->arena.alloc(plan.MK * net->esize);
            if (wantB) dstB = net->arena.alloc(plan.KN * net->esize);
            if ((wantA && !dstA) || (wantB && !dstB)) {
              // partial shortage: keep it simple, pack inline at s+1
              if (dstA) net->arena.release(dstA);
              if (dstB) net->arena.release(dstB);
              dstA = dstB = nullptr;
            }
            if (dstA || dstB) {
              hipEvent_t e_ready = nullptr, e_done = nullptr;
              if (hipEventCreate(&e_ready) != hipSuccess) e_ready = nullptr;
              if (e_ready && hipEventCreate(&e_done) != hipSuccess) {
                (void)hipEventDestroy(e_ready);
                e_ready = nullptr;
              }
              if (!e_ready) {
                if (dstA) net->arena.release(dstA);
                if (dstB) net->arena.release(dstB);
              } else {
                prep_events.push_back(e_ready);
                prep_events.push_back(e_done);
                if (hipEventRecord(e_ready, net->stream) != hipSuccess ||
                    hipStreamWaitEvent(net->stream2, e_ready, 0) !=
                        hipSuccess) {
                  rc = TN_ERR_HIP;
                  g_last_error = "prepack event sync failed";
                  break;
                }
                auto do_pack = [&](DevTensor& S, void* dst,
                                   const std::vector<int>& axes,
                                   u64 elems) -> int {
                  Meta ms;
                  meta_of(S, &ms);
                  std::vector<AxisInfo> ax;
                  for (int axis : axes)
                    ax.push_back({ms.dims[axis], ms.strides[axis], 0});
                  int prc =
                      net->dtype == 0
                          ? launch_permute((const double2*)S.data,
                                           (double2*)dst, elems, ax,
                                           net->stream2)
                          : launch_permute((const float2*)S.data,
                                           (float2*)dst, elems, ax,
                                           net->stream2);
                  if (prc != TN_OK) return prc;
                  pending.push_back({S.owned ? S.data : nullptr, S.owned});
                  std::vector<u64> nl, nd;
                  for (int axis : axes) {
                    nl.push_back(S.labels[axis]);
                    nd.push_back(S.dims[axis]);
                  }
                  S.labels = std::move(nl);
                  S.dims = std::move(nd);
                  S.data = dst;
                  S.owned = true;
                  S.external = false;
                  return TN_OK;
                };
                if (dstA) rc = do_pack(slots[ni], dstA, plan.a_axes, plan.MK);
                if (rc == TN_OK && dstB)
                  rc = do_pack(slots[nj], dstB, plan.b_axes, plan.KN);
                if (rc != TN_OK) break;
                if (hipEventRecord(e_done, net->stream2) != hipSuccess) {
                  rc = TN_ERR_HIP;
                  g_last_error = "prepack event record failed";
                  break;
                }
                pending_done = e_done;
              }
            }
          }
        }
      }
    }

    Meta ma, mb;
    meta_of(A, &ma);
    meta_of(B, &mb);

    if (profiled) HIP_CHECK(hipEventRecord(ev[4 * s], net->stream));
    StepStats st;
    if (profiled) {
      st.gemm_ev0 = ev[4 * s + 2];
      st.gemm_ev1 = ev[4 * s + 3];
    }
    if (net->dtype == 0)
      rc = einsum_dev_impl<double2>(out.labels.data(), out.dims.data(),
                                    (int)out.labels.size(), ma, mb,
                                    (double2*)out.data, net->stream, ws, &st);
    else
      rc = einsum_dev_impl<float2>(out.labels.data(), out.dims.data(),
                                   (int)out.labels.size(), ma, mb,
                                   (float2*)out.data, net->stream, ws, &st);
    if (profiled) HIP_CHECK(hipEventRecord(ev[4 * s + 1], net->stream));
    if (kind) kind[s] = st.kind;
    if (profiled) kinds_local[s] = st.kind;
    if (rc != TN_OK) {
      char buf[96];
      snprintf(buf, sizeof buf, " (at step %zu, out elems %llu)", s,
               (unsigned long long)out.elems);
      g_last_error += buf;
      ws_free(ws, out.data);
      break;
    }
    // free consumed intermediates (leaves persist)
    if (A.owned) ws_free(ws, A.data);
    if (B.owned) ws_free(ws, B.data);
    slots[i] = std::move(out);
    alive[j] = 0;
  }

  if (!capture) {
    HIP_CHECK(hipEventRecord(walk_end, net->stream));
    HIP_CHECK(hipStreamSynchronize(net->stream));
  }
  if (rc == TN_OK) {
    float ms = 0.f;
    if (!capture) {
      HIP_CHECK(hipEventElapsedTime(&ms, walk_start, walk_end));
    }
    if (elapsed_ms) *elapsed_ms = (double)ms + t0_ms;
    if (profiled) {
      for (size_t s = 0; s < nsteps; ++s) {
        float sms = 0.f;
        HIP_CHECK(hipEventElapsedTime(&sms, ev[4 * s], ev[4 * s + 1]));
        step_ms[s] = (double)sms;
        if (gemm_ms) {
          float gms = 0.f;
          // only TTGT steps (kind 2/3) record the gemm events; probing
          // unrecorded events would fail AND leave a sticky TLS error
          if ((kinds_local[s] == 2 || kinds_local[s] == 3) &&
              hipEventElapsedTime(&gms, ev[4 * s + 2], ev[4 * s + 3]) ==
                  hipSuccess)
            gemm_ms[s] = (double)gms;
          else
            gemm_ms[s] = 0.0;
        }
      }
      (void)hipGetLastError();  // clear any error from unrecorded-event reads
    }
    // locate final tensor
    int final_idx = -1;
    int count = 0;
    for (size_t x = 0; x < n; ++x)
      if (alive[x]) {
        final_idx = (int)x;
        ++count;
      }
    if (count != 1) {
      rc = TN_ERR_INVALID;
      g_last_error = "path did not fully contract the network";
    } else {
      DevTensor& f = slots[final_idx];
      if (!f.owned) {
        // final == a leaf (empty path): copy so result is stable
        DevTensor c = f;
        if (hipMalloc((void**)&c.data, f.elems * net->esize) != hipSuccess) {
          rc = TN_ERR_OOM;
          g_last_error = "hipMalloc failed for final copy";
        } else {
          HIP_CHECK(hipMemcpy(c.data, f.data, f.elems * net->esize,
                              hipMemcpyDeviceToDevice));
          c.owned = true;
          net->final_t = std::move(c);
          net->has_final = true;
        }
      } else {
        net->final_in_arena = ws.arena && ws.arena->owns(f.data);
        net->final_t = std::move(f);
        net->has_final = true;
      }
    }
  }
  if (rc != TN_OK) {
    // free any owned intermediates left over, plus originals whose
    // prepack-deferred frees never ran
    if (!capture) {
      (void)hipStreamSynchronize(net->stream);
      (void)hipStreamSynchronize(net->stream2);
    }
    for (auto& p : pending)
      if (p.old_owned && p.old_data) ws_free(ws, p.old_data);
    for (size_t x = 0; x < n; ++x)
      if (alive[x] && slots[x].owned && slots[x].data)
        ws_free(ws, slots[x].data);
  }
  if (spilled) *spilled = ws.spilled;
  if (walk_start) (void)hipEventDestroy(walk_start);
  if (walk_end) (void)hipEventDestroy(walk_end);
  for (auto& e : ev) (void)hipEventDestroy(e);
  for (auto& e : prep_events) (void)hipEventDestroy(e);
  return rc;
}

static bool graph_pairs_match(const tn_net* net, const u64* pairs,
                              size_t nsteps) {
  return net->graph_pairs.size() == 2 * nsteps &&
         std::equal(net->graph_pairs.begin(), net->graph_pairs.end(), pairs);
}

static int contract_graph_replay(tn_net* net, double* elapsed_ms) {
  HIP_CHECK(hipSetDevice(net->device));
  hipEvent_t e0 = nullptr, e1 = nullptr;
  HIP_CHECK(hipEventCreate(&e0));
  HIP_CHECK(hipEventCreate(&e1));
  HIP_CHECK(hipEventRecord(e0, net->stream));
  if (hipGraphLaunch(net->graph_exec, net->stream) != hipSuccess) {
    (void)hipGetLastError();
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    net->invalidate_graph();
    net->graph_state = -1;
    FAILV(TN_ERR_HIP, "hipGraphLaunch failed");
  }
  HIP_CHECK(hipEventRecord(e1, net->stream));
  HIP_CHECK(hipStreamSynchronize(net->stream));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
  (void)hipEventDestroy(e0);
  (void)hipEventDestroy(e1);
  if (elapsed_ms) *elapsed_ms = (double)ms;
  // bookkeeping is untouched: the replay rewrites the same arena blocks and
  // the final tensor lands at its captured address (net->final_t)
  net->has_final = true;
  return TN_OK;
}

static int contract_graph_capture(tn_net* net, const u64* pairs,
                                  size_t nsteps) {
  HIP_CHECK(hipSetDevice(net->device));
  // release the previous final OUTSIDE capture (the release may hipFree,
  // and hipFree device-syncs — illegal while the stream captures)
  if (net->has_final && net->final_t.owned) {
    if (net->final_in_arena)
      net->arena.release(net->final_t.data);
    else
      (void)hipFree(net->final_t.data);
    net->final_t = DevTensor();
    net->has_final = false;
    net->final_in_arena = false;
  }
  HIP_CHECK(hipStreamSynchronize(net->stream));
  if (hipStreamBeginCapture(net->stream, hipStreamCaptureModeThreadLocal) !=
      hipSuccess) {
    (void)hipGetLastError();
    net->graph_state = -1;
    return TN_ERR_HIP;
  }
  std::vector<void*> deferred;
  bool spilled = false;
  int rc = contract_impl(net, pairs, nsteps, nullptr, nullptr, nullptr,
                         nullptr, /*capture=*/true, &deferred, &spilled);
  hipGraph_t g = nullptr;
  hipError_t ce = hipStreamEndCapture(net->stream, &g);
  for (void* p : deferred) (void)hipFree(p);
  (void)hipGetLastError();
  if (rc != TN_OK || ce != hipSuccess || !g || spilled) {
    if (g) (void)hipGraphDestroy(g);
    net->graph_state = -1;  // caller falls back to a normal run
    return rc != TN_OK ? rc : TN_ERR_HIP;
  }
  hipGraphExec_t exec = nullptr;
  if (hipGraphInstantiate(&exec, g, nullptr, nullptr, 0) != hipSuccess) {
    (void)hipGetLastError();
    (void)hipGraphDestroy(g);
    net->graph_state = -1;
    return TN_ERR_HIP;
  }
  (void)hipGraphDestroy(g);
  net->graph_exec = exec;
  net->graph_state = 2;
  return TN_OK;
}

extern "C" int tn_net_contract(tn_net* net, const u64* pairs, size_t nsteps,
                               double* elapsed_ms) {
  if (!net) FAILV(TN_ERR_INVALID, "null net");
  if (net->graph_state == 2) {
    // a failed walk since capture clears final_t, whose address the replay
    // relies on — in that case re-run (and re-capture) normally
    if (graph_pairs_match(net, pairs, nsteps) && net->final_t.data)
      return contract_graph_replay(net, elapsed_ms);
    net->invalidate_graph();  // different path / lost state: re-arm below
  }
  if (net->graph_state == 1 && net->arena.base &&
      graph_pairs_match(net, pairs, nsteps)) {
    if (contract_graph_capture(net, pairs, nsteps) == TN_OK)
      return contract_graph_replay(net, elapsed_ms);
    // capture failed (workspace spill etc.) — run normally below
  }
  int rc = contract_impl(net, pairs, nsteps, nullptr, nullptr, nullptr,
                         elapsed_ms);
  if (rc == TN_OK && net->graph_state >= 0 && net->arena.base) {
    net->graph_pairs.assign(pairs, pairs + 2 * nsteps);
    net->graph_state = 1;  // arm capture for the next identical call
  }
  return rc;
}

extern "C" int tn_net_contract_profiled(tn_net* net, const u64* pairs,
                                        size_t nsteps, double* step_ms,
                                        double* gemm_ms, int32_t* kind,
                                        double* elapsed_ms) {
  if (!net) FAILV(TN_ERR_INVALID, "null net");
  if (net->graph_state == 2 && !graph_pairs_match(net, pairs, nsteps))
    net->invalidate_graph();  // different path moves the final block
  int rc =
      contract_impl(net, pairs, nsteps, step_ms, gemm_ms, kind, elapsed_ms);
  if (rc == TN_OK && net->graph_state == 0 && net->arena.base) {
    net->graph_pairs.assign(pairs, pairs + 2 * nsteps);
    net->graph_state = 1;  // a profiled pass also arms graph capture
  }
  return rc;
}

extern "C" int tn_memcpy_dtod(void* dst, const void* src, u64 bytes) {
  HIP_CHECK(hipMemcpy(dst, src, bytes, hipMemcpyDeviceToDevice));
  return TN_OK;
}

extern "C" int tn_memcpy_dtoh(void* host_dst, const void* dev_src,
                              u64 bytes) {
  HIP_CHECK(hipMemcpy(host_dst, dev_src, bytes, hipMemcpyDeviceToHost));
  return TN_OK;
}

extern "C" int tn_net_result_meta(tn_net* net, u64* labels, u64* dims,
                                  size_t* ndim) {
  if (!net || !net->has_final) FAILV(TN_ERR_INVALID, "no result available");
  size_t nd = net->final_t.labels.size();
  *ndim = nd;
  for (size_t i = 0; i < nd; ++i) {
    labels[i] = net->final_t.labels[i];
    dims[i] = net->final_t.dims[i];
  }
  return TN_OK;
}

extern "C" int tn_net_result_data(tn_net* net, void* host_out) {
  if (!net || !net->has_final) FAILV(TN_ERR_INVALID, "no result available");
  HIP_CHECK(hipSetDevice(net->device));
  HIP_CHECK(hipMemcpy(host_out, net->final_t.data,
                      net->final_t.elems * net->esize,
                      hipMemcpyDeviceToHost));
  return TN_OK;
}

extern "C" void* tn_net_result_dev(tn_net* net) {
  if (!net || !net->has_final) return nullptr;
  return net->final_t.data;
}

extern "C" int tn_net_pool_bytes(tn_net* net, u64* in_use, u64* cached) {
  if (!net) FAILV(TN_ERR_INVALID, "null net");
  // arena telemetry (the stream-ordered pool is unused — see ws_alloc)
  u64 used = 0;
  for (const auto& b : net->arena.blocks)
    if (!b.free) used += b.sz;
  if (in_use) *in_use = used;
  if (cached) *cached = net->arena.size;
  return TN_OK;
}

extern "C" void tn_net_destroy(tn_net* net) {
  if (!net) return;
  (void)hipSetDevice(net->device);
  (void)hipStreamSynchronize(net->stream);
  net->invalidate_graph();
  for (auto& t : net->leaves)
    if (t.data && !t.owned && !t.external) (void)hipFree(t.data);
  if (net->has_final && net->final_t.owned && !net->final_in_arena)
    (void)hipFree(net->final_t.data);
  net->arena.destroy();
  (void)hipStreamDestroy(net->stream);
  if (net->stream2) (void)hipStreamDestroy(net->stream2);
  delete net;
}
