// blades_amd CDNA4 (gfx950 / MI355X) kernel library.
//
// Hand-written HIP kernels for the robust-aggregation / attack hot path
// (SURVEY.md §2.4 K1-K11).  All kernels operate on the stacked update
// matrix U ∈ R^{K×d} (fp32, row stride ld >= d), which the runtime keeps
// resident in HBM3E.  Design notes per kernel at the definition.
//
// Conventions:
//  * wavefront = 64 (CDNA), block sizes are multiples of 64
//  * memory-bound kernels: float4 loads (16 B/lane) over the vectorizable
//    column span, scalar tail; grid capped ~2048 blocks + grid-stride
//  * the Gram kernel uses the f32 MFMA (v_mfma_f32_32x32x2_f32) — exact
//    fp32 numerics at the 157 TF f32 rate, LDS-staged tiles
//
// Build: hipcc --offload-arch=gfx950 via torch.utils.cpp_extension (see
// setup.py).  No CUDA shims, no hipify — this file is native HIP.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <tuple>

#define CHECK_IN(x)                                                     \
  TORCH_CHECK((x).is_cuda(), #x " must be on GPU");                     \
  TORCH_CHECK((x).scalar_type() == at::kFloat, #x " must be fp32");

namespace {

constexpr int kMaxBlocks = 4096;  // 256 CU x ~16 blocks

__host__ __device__ inline long long cdiv(long long a, long long b) {
  return (a + b - 1) / b;
}

// ---------------------------------------------------------------------------
// Column reductions (K1, GeoMed/FLTrust accumulation, ALIE/IPM stats)
// ---------------------------------------------------------------------------
// One thread owns one (vector of 4) column(s); lane i and lane i+1 touch
// adjacent float4s -> fully coalesced 16 B/lane streams.  The K-loop walks
// rows at stride ld.  MODE selects the accumulation:
//   0: mean            out0[j] = (1/K) sum_k U[k][j]
//   1: weighted sum    out0[j] = sum_k w[k] U[k][j]
//   2: masked mean     out0[j] = (1/|M|) sum_{k in M} U[k][j]
//   3: masked mean+var out0=mean, out1=std over mask (Welford-free two-sum:
//      fp32 sum + sumsq is adequate at K<=1e4 given |U| ~ lr*grad scales;
//      matches torch.std to ~1e-6 in tests)

template <int MODE, bool VEC>
__global__ void col_reduce_kernel(const float* __restrict__ U,
                                  const float* __restrict__ w,
                                  const bool* __restrict__ mask,
                                  float* __restrict__ out0,
                                  float* __restrict__ out1,
                                  long long K, long long d, long long ld,
                                  float inv_count, bool unbiased,
                                  float count) {
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;

  if (VEC) {
    const long long d4 = d / 4;
    for (long long j4 = tid; j4 < d4; j4 += nthreads) {
      float4 acc = make_float4(0.f, 0.f, 0.f, 0.f);
      float4 acc2 = make_float4(0.f, 0.f, 0.f, 0.f);
      const float* col = U + 4 * j4;
      for (long long k = 0; k < K; ++k) {
        if (MODE == 2 || MODE == 3) {
          if (!mask[k]) continue;
        }
        const float4 v = *reinterpret_cast<const float4*>(col + k * ld);
        const float wk = (MODE == 1) ? w[k] : 1.f;
        acc.x += wk * v.x; acc.y += wk * v.y;
        acc.z += wk * v.z; acc.w += wk * v.w;
        if (MODE == 3) {
          acc2.x += v.x * v.x; acc2.y += v.y * v.y;
          acc2.z += v.z * v.z; acc2.w += v.w * v.w;
        }
      }
      if (MODE == 0 || MODE == 2) {
        acc.x *= inv_count; acc.y *= inv_count;
        acc.z *= inv_count; acc.w *= inv_count;
      }
      *reinterpret_cast<float4*>(out0 + 4 * j4) = acc;
      if (MODE == 3) {
        const float denom = unbiased ? (count - 1.f) : count;
        float4 sd;
        sd.x = sqrtf(fmaxf((acc2.x - acc.x * acc.x / count), 0.f) / denom);
        sd.y = sqrtf(fmaxf((acc2.y - acc.y * acc.y / count), 0.f) / denom);
        sd.z = sqrtf(fmaxf((acc2.z - acc.z * acc.z / count), 0.f) / denom);
        sd.w = sqrtf(fmaxf((acc2.w - acc.w * acc.w / count), 0.f) / denom);
        *reinterpret_cast<float4*>(out1 + 4 * j4) = sd;
        // mean output for MODE 3
        float4 mu;
        mu.x = acc.x / count; mu.y = acc.y / count;
        mu.z = acc.z / count; mu.w = acc.w / count;
        *reinterpret_cast<float4*>(out0 + 4 * j4) = mu;
      }
    }
  }

  // scalar span: the whole matrix when !VEC, else just the tail columns
  const long long j_begin = VEC ? (d / 4) * 4 : 0;
  for (long long j = j_begin + tid; j < d; j += nthreads) {
    float acc = 0.f, acc2 = 0.f;
    for (long long k = 0; k < K; ++k) {
      if (MODE == 2 || MODE == 3) {
        if (!mask[k]) continue;
      }
      const float v = U[k * ld + j];
      const float wk = (MODE == 1) ? w[k] : 1.f;
      acc += wk * v;
      if (MODE == 3) acc2 += v * v;
    }
    if (MODE == 0 || MODE == 2) acc *= inv_count;
    if (MODE == 3) {
      const float denom = unbiased ? (count - 1.f) : count;
      out1[j] = sqrtf(fmaxf((acc2 - acc * acc / count), 0.f) / denom);
      out0[j] = acc / count;
    } else {
      out0[j] = acc;
    }
  }
}

// ---------------------------------------------------------------------------
// K3/K2 — coordinate-wise trimmed sum / median selection
// ---------------------------------------------------------------------------
// One thread owns one coordinate and streams its column (coalesced across
// lanes), maintaining the b smallest and b largest values seen so far in
// per-thread LDS scratch.  Expected replacement work is b·ln(K/b) per
// column, so the kernel stays memory-bound for the target shapes
// (K=100, b=20 headline config).  Output: trimmed mean over the middle
// K-2b values.  Median = b=(K-1)/2 (averages both middles for even K,
// matching reference semantics, aggregators/median.py:23-25).
//
// LDS layout: per-thread strided (lo[i*BS + tid]) so each 32-lane group
// hits 32 distinct banks (b32 banking).  NaNs must be sanitized upstream
// (get_update applies nan_to_num — client.py:198 semantics).

// Per-thread binary heaps in LDS (strided layout): lo = MAX-heap holding
// the b smallest seen, hi = MIN-heap holding the b largest.  The heap
// roots are cached in registers, so the streaming common case (value not
// entering either buffer) costs two register compares; a replacement
// costs one O(log b) sift-down in LDS (the previous scan-based
// maintenance cost O(b) per replacement and measured 7.5 ms at K=100,
// b=20 — 10x the memory floor).

__device__ __forceinline__ void sift_down_max(float* h, int BS, int b,
                                              float v) {
  int i = 0;
  for (;;) {
    const int l = 2 * i + 1, r = 2 * i + 2;
    int big = i;
    float bv = v;
    if (l < b) { const float x = h[l * BS]; if (x > bv) { big = l; bv = x; } }
    if (r < b) { const float x = h[r * BS]; if (x > bv) { big = r; bv = x; } }
    if (big == i) break;
    h[i * BS] = bv;
    i = big;
  }
  h[i * BS] = v;
}

__device__ __forceinline__ void sift_down_min(float* h, int BS, int b,
                                              float v) {
  int i = 0;
  for (;;) {
    const int l = 2 * i + 1, r = 2 * i + 2;
    int sm = i;
    float sv = v;
    if (l < b) { const float x = h[l * BS]; if (x < sv) { sm = l; sv = x; } }
    if (r < b) { const float x = h[r * BS]; if (x < sv) { sm = r; sv = x; } }
    if (sm == i) break;
    h[i * BS] = sv;
    i = sm;
  }
  h[i * BS] = v;
}

__global__ void trimmed_select_kernel(const float* __restrict__ U,
                                      float* __restrict__ out,
                                      long long K, long long d, long long ld,
                                      int b) {
  extern __shared__ float smem[];
  const int BS = blockDim.x;
  float* lo = smem + threadIdx.x;            // b floats, stride BS
  float* hi = smem + (size_t)b * BS + threadIdx.x;

  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const float inv = 1.f / (float)(K - 2 * b);

  for (long long j = tid; j < d; j += nthreads) {
    // fp64 running/trim sums: the result is (sum − trim) where the two can
    // cancel to ~(K−2b)/K of their magnitude (median: 2 of K survive) —
    // fp32 here loses ~1e-6 absolute; fp64 makes the subtraction exact at
    // fp32 output precision.
    double sum = 0.0;
    float lo_root = -INFINITY, hi_root = INFINITY;

    // fill phase: the first b values seed both heaps (sift-up inserts)
    long long k = 0;
    for (; k < (b < K ? b : K); ++k) {
      const float v = U[k * ld + j];
      sum += v;
      int i = (int)k;  // insert at the end of the max-heap, sift up
      while (i > 0) {
        const int p = (i - 1) / 2;
        const float pv = lo[p * BS];
        if (pv >= v) break;
        lo[i * BS] = pv;
        i = p;
      }
      lo[i * BS] = v;
      i = (int)k;      // min-heap
      while (i > 0) {
        const int p = (i - 1) / 2;
        const float pv = hi[p * BS];
        if (pv <= v) break;
        hi[i * BS] = pv;
        i = p;
      }
      hi[i * BS] = v;
    }
    if (b > 0 && K >= b) {
      lo_root = lo[0];
      hi_root = hi[0];
    }

    // stream phase: register compare; sift only on replacement.  8-wide
    // unroll issues independent loads together — the 4 B/lane column
    // stream is latency-bound otherwise (measured 8.7 ms vs the 0.7 ms
    // HBM floor at K=100, d=11.2M without it).
    if (k + 8 <= K) {
      // software pipeline: group g+1's loads issue BEFORE group g's
      // branchy heap maintenance (hipcc otherwise fences loads behind the
      // data-dependent LDS branches — guide §5.4 rule 4c)
      float v8[8];
      #pragma unroll
      for (int u = 0; u < 8; ++u) v8[u] = U[(k + u) * ld + j];
      k += 8;
      for (; k + 8 <= K; k += 8) {
        float n8[8];
        #pragma unroll
        for (int u = 0; u < 8; ++u) n8[u] = U[(k + u) * ld + j];
        #pragma unroll
        for (int u = 0; u < 8; ++u) {
          const float v = v8[u];
          sum += v;
          // non-divergent maintenance: when ANY lane replaces, every lane
          // sifts — non-replacing lanes sift the current root back in (an
          // identity: the root is the extremum, so it settles at slot 0
          // immediately).  The divergent form serialized a sparse sift on
          // nearly every row (7.5 ms vs 0.7 ms floor at K=100, b=20).
          if (__any(v < lo_root)) {
            sift_down_max(lo, BS, b, (v < lo_root) ? v : lo_root);
            lo_root = lo[0];
          }
          if (__any(v > hi_root)) {
            sift_down_min(hi, BS, b, (v > hi_root) ? v : hi_root);
            hi_root = hi[0];
          }
        }
        #pragma unroll
        for (int u = 0; u < 8; ++u) v8[u] = n8[u];
      }
      #pragma unroll
      for (int u = 0; u < 8; ++u) {
        const float v = v8[u];
        sum += v;
        if (v < lo_root) {
          sift_down_max(lo, BS, b, v);
          lo_root = lo[0];
        }
        if (v > hi_root) {
          sift_down_min(hi, BS, b, v);
          hi_root = hi[0];
        }
      }
    }
    for (; k < K; ++k) {
      const float v = U[k * ld + j];
      sum += v;
      if (v < lo_root) {
        sift_down_max(lo, BS, b, v);
        lo_root = lo[0];
      }
      if (v > hi_root) {
        sift_down_min(hi, BS, b, v);
        hi_root = hi[0];
      }
    }
    double trim = 0.0;
    for (int i = 0; i < b; ++i)
      trim += (double)lo[i * BS] + (double)hi[i * BS];
    out[j] = (float)((sum - trim) * (double)inv);
  }
}

// ---------------------------------------------------------------------------
// K3/K2 small-b fast path — divergence-free register selection
// ---------------------------------------------------------------------------
// The LDS-heap kernel above is divergence-bound at the headline shape
// (K=100, b=20): with 64 lanes and b/K = 0.2, some lane replaces on nearly
// every row, so the ballot-gated sift serializes the wave ~every step
// (measured 7.5 ms vs the 0.7 ms HBM floor; PMC showed SQ busy ~12%,
// profiles/r01_pmc_aggregation.csv).  This kernel has NO data-dependent
// control flow in the hot path at all:
//
//   * one coordinate per LANE; each lane keeps the NB smallest (lo, sorted
//     ascending) and NB largest (hi, sorted descending) values seen, in
//     REGISTERS with static indices (b <= NB <= 32);
//   * rows stream in groups of NB: one in-register bitonic sort of the
//     group (min/max compare-exchange network, fully unrolled), then a
//     bitonic lower-half merge lo[i] <-> g[NB-1-i] + 5-stage clean — every
//     lane executes the identical instruction stream;
//   * a wave ballot gates the sort+merge on "any lane's group crosses its
//     thresholds": always true in the first rows, vanishingly rare once
//     K >> b (expected triggered groups ~ 64·NB·ln(K)/K of rows), so the
//     kernel degrades to a pure float-stream + f64 accumulate at large K;
//   * fp64 running sum (cancellation under Byzantine-magnitude outliers:
//     result = sum − trims; see commit 7977a4c).
//
// VALU cost per value when every group triggers (K ~ 100): bitonic sort
// ~240 CE + 2 merges ~112 CE each = ~930 min/max ops per NB=32 group
// = ~29/value -> ~0.45 ms of VALU at K=100, d=11.2M — under the 0.71 ms
// HBM floor, i.e. memory-bound by construction.

template <int NB>
__device__ __forceinline__ void bitonic_sort_asc(float (&g)[NB]) {
  #pragma unroll
  for (int k = 2; k <= NB; k <<= 1) {
    #pragma unroll
    for (int j = k >> 1; j > 0; j >>= 1) {
      #pragma unroll
      for (int i = 0; i < NB; ++i) {
        const int l = i ^ j;
        if (l > i) {
          const bool up = ((i & k) == 0);
          const float a = g[i], b = g[l];
          g[i] = up ? fminf(a, b) : fmaxf(a, b);
          g[l] = up ? fmaxf(a, b) : fminf(a, b);
        }
      }
    }
  }
}

// clean a bitonic sequence into ascending (ASC) or descending order
template <int NB, bool ASC>
__device__ __forceinline__ void bitonic_clean(float (&m)[NB]) {
  #pragma unroll
  for (int j = NB >> 1; j > 0; j >>= 1) {
    #pragma unroll
    for (int i = 0; i < NB; ++i) {
      const int l = i ^ j;
      if (l > i) {
        const float a = m[i], b = m[l];
        m[i] = ASC ? fminf(a, b) : fmaxf(a, b);
        m[l] = ASC ? fmaxf(a, b) : fminf(a, b);
      }
    }
  }
}

// b_lo / b_hi: drop that many smallest / largest values (either may be 0;
// both <= NB).  inv scales the trimmed sum: 1/(K-b_lo-b_hi) for the
// trimmed mean, 1.0 for raw trimmed sums (K5 krum scores: the score of
// column i of the symmetric zero-diagonal distance matrix is the sum of
// its n-f-2 smallest off-diagonal entries = col_trimmed_sum(D, 0, f+1),
// reference: aggregators/krum.py:9-25).
template <int NB>
__global__ __launch_bounds__(256)
void trimmed_regsel_kernel(const float* __restrict__ U,
                           float* __restrict__ out,
                           long long K, long long d, long long ld,
                           int b_lo, int b_hi, double inv) {
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;

  for (long long j = tid; j < d; j += nthreads) {
    float lo[NB], hi[NB], g[NB];
    #pragma unroll
    for (int i = 0; i < NB; ++i) { lo[i] = INFINITY; hi[i] = -INFINITY; }
    // 4 partial f64 accumulators break the serial add chain in the
    // unrolled group loop (dependent v_add_f64 latency would serialize)
    double s0 = 0.0, s1 = 0.0, s2 = 0.0, s3 = 0.0;

    long long k = 0;
    for (; k + NB <= K; k += NB) {
      // group load + f64 accumulate + threshold test (branch-free)
      bool need = false;
      #pragma unroll
      for (int u = 0; u < NB; ++u) {
        const float v = U[(k + u) * ld + j];
        g[u] = v;
        if ((u & 3) == 0) s0 += (double)v;
        else if ((u & 3) == 1) s1 += (double)v;
        else if ((u & 3) == 2) s2 += (double)v;
        else s3 += (double)v;
        need |= (v < lo[NB - 1]) | (v > hi[NB - 1]);
      }
      if (__any(need)) {
        bitonic_sort_asc<NB>(g);
        // half bitonic merges IN PLACE (lo/hi are overwritten by the
        // merge result anyway — no temporaries, ~96 regs peak).
        // lo (ascending) pairs with reversed g; hi (descending) pairs
        // with g directly: by the negation duality (-hi ascending,
        // reversed(-g) ascending) the lower-half cleaner min(-hi[i],
        // -g[NB-1-(NB-1-i)]) = -max(hi[i], g[i]).
        #pragma unroll
        for (int i = 0; i < NB; ++i) lo[i] = fminf(lo[i], g[NB - 1 - i]);
        bitonic_clean<NB, true>(lo);
        #pragma unroll
        for (int i = 0; i < NB; ++i) hi[i] = fmaxf(hi[i], g[i]);
        bitonic_clean<NB, false>(hi);
      }
    }
    // tail rows: branchless bubble insert (static indices)
    for (; k < K; ++k) {
      const float v = U[k * ld + j];
      s0 += (double)v;
      float c = v;
      #pragma unroll
      for (int i = 0; i < NB; ++i) {
        const float a = lo[i];
        lo[i] = fminf(a, c);
        c = fmaxf(a, c);
      }
      c = v;
      #pragma unroll
      for (int i = 0; i < NB; ++i) {
        const float a = hi[i];
        hi[i] = fmaxf(a, c);
        c = fminf(a, c);
      }
    }
    // predicated static-index sum — a runtime-bound loop would force the
    // register arrays to scratch (dynamic indexing)
    double trim = 0.0;
    #pragma unroll
    for (int i = 0; i < NB; ++i) {
      trim += (i < b_lo) ? (double)lo[i] : 0.0;
      trim += (i < b_hi) ? (double)hi[i] : 0.0;
    }
    out[j] = (float)(((s0 + s1) + (s2 + s3) - trim) * inv);
  }
}

// ---------------------------------------------------------------------------
// K2/K3 at large K — dual radix-select trimmed mean
// ---------------------------------------------------------------------------
// The LDS streaming-selection kernel above needs 2·b floats of LDS per
// thread, capping b ≈ 127.  Config 3/5 scales (K = 1e3..1e4, median ⇒
// b = (K−1)/2) use this kernel instead: per coordinate, find BOTH trim
// thresholds (the b-th and (K−b−1)-th order statistics) by 4 levels of
// byte-wise MSB radix selection on the order-preserving fp32→u32 key, then
// one final pass sums the kept middle band with exact tie handling.
// 5 passes over the slab total; each block owns a 32-coordinate tile with
// two [32][256] u32 histograms in LDS (atomic bin increments), per-tile
// element reads fully coalesced (consecutive lanes → consecutive columns).

__device__ __forceinline__ unsigned int order_key(float x) {
  unsigned int u = __float_as_uint(x);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

__device__ __forceinline__ float key_value(unsigned int key) {
  unsigned int u = (key & 0x80000000u) ? (key & 0x7fffffffu) : ~key;
  return __uint_as_float(u);
}

// --------------------------------------------------------------------------
// LDS-staged dual radix select — K <= 1280 (config 3 scale)
// --------------------------------------------------------------------------
// The 5-pass kernel below re-reads the whole K×d slab per level because
// per-coordinate histograms cap the tile at 32 coordinates; at config 3
// (K=1000, median, d=11.2M) that is 5×44.7 GB of HBM traffic (measured
// 206 ms).  Here the [K, 16] tile is staged into LDS ONCE and every
// radix level plus the final band pass sweep LDS instead — exactly one
// HBM read of the slab.  LDS budget at K=1280: tile 80 KiB + two
// [16][256] histograms 32 KiB + small arrays ≈ 113 KiB (dynamic LDS,
// 160 KiB/CU ceiling; one workgroup per CU).
constexpr int RT_L = 16;   // coordinates per LDS-staged tile
constexpr long long RT_L_KMAX = 1280;

__global__ __launch_bounds__(256)
void radix_trimmed_lds_kernel(const float* __restrict__ U,
                              float* __restrict__ out,
                              long long K, long long d, long long ld,
                              long b_lo, long b_hi, double inv) {
  extern __shared__ unsigned char smem_raw[];
  // doubles first (8-aligned at offset 0), then u32 arrays, then the tile
  double* fsum = reinterpret_cast<double*>(smem_raw);
  unsigned int* u32base = reinterpret_cast<unsigned int*>(fsum + RT_L);
  unsigned int* feqA = u32base;
  unsigned int* feqB = feqA + RT_L;
  unsigned int* prefA = feqB + RT_L;
  unsigned int* baseA = prefA + RT_L;
  unsigned int* prefB = baseA + RT_L;
  unsigned int* baseB = prefB + RT_L;
  unsigned int* histA = baseB + RT_L;          // [RT_L][256]
  unsigned int* histB = histA + RT_L * 256;
  float* tile = reinterpret_cast<float*>(histB + RT_L * 256);  // [K][RT_L]

  const long long kA = b_lo;
  const long long kB = K - b_hi - 1;
  const long long n_el = K * RT_L;

  for (long long j0 = (long long)blockIdx.x * RT_L; j0 < d;
       j0 += (long long)gridDim.x * RT_L) {
    const int tw = (int)((d - j0) < RT_L ? (d - j0) : RT_L);
    const bool vec = (ld % 4 == 0) && (j0 + RT_L <= d);
    // ------------------------------------------------ stage tile to LDS
    __syncthreads();  // previous iteration's sweeps are done
    if (vec) {
      for (long long e = threadIdx.x; e < K * (RT_L / 4);
           e += blockDim.x) {
        const int t4 = (int)(e % (RT_L / 4));
        const long long k = e / (RT_L / 4);
        const float4 v = *reinterpret_cast<const float4*>(
            U + k * ld + j0 + 4 * t4);
        float* dst = tile + k * RT_L + 4 * t4;
        dst[0] = v.x; dst[1] = v.y; dst[2] = v.z; dst[3] = v.w;
      }
    } else {
      for (long long e = threadIdx.x; e < n_el; e += blockDim.x) {
        const int t = (int)(e % RT_L);
        const long long k = e / RT_L;
        tile[k * RT_L + t] = (t < tw) ? U[k * ld + j0 + t] : 0.f;
      }
    }
    if (threadIdx.x < RT_L) {
      prefA[threadIdx.x] = 0; baseA[threadIdx.x] = 0;
      prefB[threadIdx.x] = 0; baseB[threadIdx.x] = 0;
    }
    // ------------------------------------------------------ radix levels
    for (int level = 3; level >= 0; --level) {
      __syncthreads();
      for (int e = threadIdx.x; e < RT_L * 256; e += blockDim.x) {
        histA[e] = 0;
        histB[e] = 0;
      }
      __syncthreads();
      const int shift = 8 * level;
      for (long long e = threadIdx.x; e < n_el; e += blockDim.x) {
        const int t = (int)(e % RT_L);
        const unsigned int key = order_key(tile[e]);
        const unsigned int hi = (level == 3) ? 0u : (key >> (shift + 8));
        const unsigned int bin = (key >> shift) & 255u;
        if (hi == prefA[t]) atomicAdd(&histA[t * 256 + bin], 1u);
        if (hi == prefB[t]) atomicAdd(&histB[t * 256 + bin], 1u);
      }
      __syncthreads();
      if (threadIdx.x < (unsigned)RT_L) {
        const int t = threadIdx.x;
        unsigned long long cum = baseA[t];
        for (int bin = 0; bin < 256; ++bin) {
          const unsigned int c = histA[t * 256 + bin];
          if (cum + c > (unsigned long long)kA) {
            prefA[t] = (prefA[t] << 8) | (unsigned)bin;
            baseA[t] = (unsigned)cum;
            break;
          }
          cum += c;
        }
        cum = baseB[t];
        for (int bin = 0; bin < 256; ++bin) {
          const unsigned int c = histB[t * 256 + bin];
          if (cum + c > (unsigned long long)kB) {
            prefB[t] = (prefB[t] << 8) | (unsigned)bin;
            baseB[t] = (unsigned)cum;
            break;
          }
          cum += c;
        }
      }
    }
    __syncthreads();
    // --------------------------------------------------- final band pass
    if (threadIdx.x < RT_L) {
      fsum[threadIdx.x] = 0.0;
      feqA[threadIdx.x] = 0;
      feqB[threadIdx.x] = 0;
    }
    __syncthreads();
    for (long long e = threadIdx.x; e < n_el; e += blockDim.x) {
      const int t = (int)(e % RT_L);
      const float x = tile[e];
      const unsigned int key = order_key(x);
      if (key > prefA[t] && key < prefB[t]) atomicAdd(&fsum[t], (double)x);
      if (key == prefA[t]) atomicAdd(&feqA[t], 1u);
      if (key == prefB[t]) atomicAdd(&feqB[t], 1u);
    }
    __syncthreads();
    if (threadIdx.x < (unsigned)tw) {
      const int t = threadIdx.x;
      const double s = fsum[t];
      const long long ea = feqA[t], eb = feqB[t];
      const double vA = (double)key_value(prefA[t]);
      const double vB = (double)key_value(prefB[t]);
      const long long cltA = baseA[t], cltB = baseB[t];
      double total;
      if (prefA[t] == prefB[t]) {
        total = (double)(K - b_lo - b_hi) * vA;
      } else {
        const long long incA =
            (cltA + ea < K - b_hi ? cltA + ea : K - b_hi) - b_lo;
        const long long incB = (K - b_hi) - (cltB > b_lo ? cltB : b_lo);
        total = s + (double)incA * vA + (double)incB * vB;
      }
      out[j0 + t] = (float)(total * inv);
    }
  }
}

constexpr int RT_T = 32;  // coordinates per block tile

__global__ __launch_bounds__(256)
void radix_trimmed_kernel(const float* __restrict__ U,
                          float* __restrict__ out,
                          long long K, long long d, long long ld,
                          long b_lo, long b_hi, double inv) {
  __shared__ unsigned int histA[RT_T][256];
  __shared__ unsigned int histB[RT_T][256];
  __shared__ unsigned int prefA[RT_T], baseA[RT_T];
  __shared__ unsigned int prefB[RT_T], baseB[RT_T];
  __shared__ double fsum[RT_T];            // final-pass per-coord sums
  __shared__ unsigned int feqA[RT_T], feqB[RT_T];

  const long long kA = b_lo;          // first kept rank (0-indexed)
  const long long kB = K - b_hi - 1;  // last kept rank

  for (long long j0 = (long long)blockIdx.x * RT_T; j0 < d;
       j0 += (long long)gridDim.x * RT_T) {
    const int tw = (int)((d - j0) < RT_T ? (d - j0) : RT_T);
    if (threadIdx.x < RT_T) {
      prefA[threadIdx.x] = 0; baseA[threadIdx.x] = 0;
      prefB[threadIdx.x] = 0; baseB[threadIdx.x] = 0;
    }
    for (int level = 3; level >= 0; --level) {
      __syncthreads();
      for (int e = threadIdx.x; e < RT_T * 256; e += blockDim.x) {
        (&histA[0][0])[e] = 0;
        (&histB[0][0])[e] = 0;
      }
      __syncthreads();
      const int shift = 8 * level;
      // float4 loads where legal (16B-aligned full tile): 1 KiB per wave
      // instruction instead of 128 B — the scalar form is latency-bound
      // (measured 712 GB/s; vectorized reads recover the HBM stream)
      const bool vec = (ld % 4 == 0) && (j0 + RT_T <= d);
      if (vec) {
        for (long long e = threadIdx.x; e < K * (RT_T / 4);
             e += blockDim.x) {
          const int t4 = (int)(e % (RT_T / 4));
          const long long k = e / (RT_T / 4);
          const float4 v = *reinterpret_cast<const float4*>(
              U + k * ld + j0 + 4 * t4);
          #pragma unroll
          for (int i = 0; i < 4; ++i) {
            const int t = 4 * t4 + i;
            const float x = (i == 0) ? v.x : (i == 1) ? v.y
                           : (i == 2) ? v.z : v.w;
            const unsigned int key = order_key(x);
            const unsigned int hi = (level == 3) ? 0u : (key >> (shift + 8));
            const unsigned int bin = (key >> shift) & 255u;
            if (hi == prefA[t]) atomicAdd(&histA[t][bin], 1u);
            if (hi == prefB[t]) atomicAdd(&histB[t][bin], 1u);
          }
        }
      } else {
        for (long long e = threadIdx.x; e < K * RT_T; e += blockDim.x) {
          const int t = (int)(e % RT_T);
          if (t >= tw) continue;
          const long long k = e / RT_T;
          const unsigned int key = order_key(U[k * ld + j0 + t]);
          const unsigned int hi = (level == 3) ? 0u : (key >> (shift + 8));
          const unsigned int bin = (key >> shift) & 255u;
          if (hi == prefA[t]) atomicAdd(&histA[t][bin], 1u);
          if (hi == prefB[t]) atomicAdd(&histB[t][bin], 1u);
        }
      }
      __syncthreads();
      if (threadIdx.x < (unsigned)tw) {
        const int t = threadIdx.x;
        unsigned long long cum = baseA[t];
        for (int bin = 0; bin < 256; ++bin) {
          const unsigned int c = histA[t][bin];
          if (cum + c > (unsigned long long)kA) {
            prefA[t] = (prefA[t] << 8) | (unsigned)bin;
            baseA[t] = (unsigned)cum;
            break;
          }
          cum += c;
        }
        cum = baseB[t];
        for (int bin = 0; bin < 256; ++bin) {
          const unsigned int c = histB[t][bin];
          if (cum + c > (unsigned long long)kB) {
            prefB[t] = (prefB[t] << 8) | (unsigned)bin;
            baseB[t] = (unsigned)cum;
            break;
          }
          cum += c;
        }
      }
    }
    __syncthreads();
    // final pass: coalesced float4 sweep; kept values accumulate into
    // per-coordinate LDS doubles (ds_add_f64), ties into u32 counters.
    // The accumulate branch fires on (K-2b)/K of elements, so the common
    // case is compare-only.
    if (threadIdx.x < RT_T) {
      fsum[threadIdx.x] = 0.0;
      feqA[threadIdx.x] = 0;
      feqB[threadIdx.x] = 0;
    }
    __syncthreads();
    {
      const bool vec = (ld % 4 == 0) && (j0 + RT_T <= d);
      if (vec) {
        for (long long e = threadIdx.x; e < K * (RT_T / 4);
             e += blockDim.x) {
          const int t4 = (int)(e % (RT_T / 4));
          const long long k = e / (RT_T / 4);
          const float4 v = *reinterpret_cast<const float4*>(
              U + k * ld + j0 + 4 * t4);
          #pragma unroll
          for (int i = 0; i < 4; ++i) {
            const int t = 4 * t4 + i;
            const float x = (i == 0) ? v.x : (i == 1) ? v.y
                           : (i == 2) ? v.z : v.w;
            const unsigned int key = order_key(x);
            if (key > prefA[t] && key < prefB[t])
              atomicAdd(&fsum[t], (double)x);
            if (key == prefA[t]) atomicAdd(&feqA[t], 1u);
            if (key == prefB[t]) atomicAdd(&feqB[t], 1u);
          }
        }
      } else {
        for (long long e = threadIdx.x; e < K * RT_T; e += blockDim.x) {
          const int t = (int)(e % RT_T);
          if (t >= tw) continue;
          const long long k = e / RT_T;
          const float x = U[k * ld + j0 + t];
          const unsigned int key = order_key(x);
          if (key > prefA[t] && key < prefB[t])
            atomicAdd(&fsum[t], (double)x);
          if (key == prefA[t]) atomicAdd(&feqA[t], 1u);
          if (key == prefB[t]) atomicAdd(&feqB[t], 1u);
        }
      }
    }
    __syncthreads();
    if (threadIdx.x < (unsigned)tw) {
      const int t = threadIdx.x;
      const double s = fsum[t];
      const long long ea = feqA[t], eb = feqB[t];
      const double vA = (double)key_value(prefA[t]);
      const double vB = (double)key_value(prefB[t]);
      const long long cltA = baseA[t], cltB = baseB[t];
      double total;
      if (prefA[t] == prefB[t]) {
        total = (double)(K - b_lo - b_hi) * vA;
      } else {
        // kept ranks are [b_lo, K-b_hi); ties at the thresholds contribute
        // the overlap of their rank range with the kept band
        const long long incA =
            (cltA + ea < K - b_hi ? cltA + ea : K - b_hi) - b_lo;
        const long long incB = (K - b_hi) - (cltB > b_lo ? cltB : b_lo);
        total = s + (double)incA * vA + (double)incB * vB;
      }
      out[j0 + t] = (float)(total * inv);
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Row reductions (K6 Weiszfeld distances, K7 clip norms, K9 FLTrust dots)
// ---------------------------------------------------------------------------
// grid = (splits, K): block (s, k) reduces slice s of row k with float4
// loads + wave shuffle + one atomicAdd per block (out zero-initialized).
//   MODE 0: sum U[k][j]^2        MODE 1: sum (U[k][j]-z[j])^2
//   MODE 2: sum U[k][j]*v[j]

template <int MODE, bool VEC>
__global__ void row_reduce_kernel(const float* __restrict__ U,
                                  const float* __restrict__ z,
                                  float* __restrict__ out,
                                  long long K, long long d, long long ld) {
  const long long k = blockIdx.y;
  const float* row = U + k * ld;
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;

  float acc = 0.f;
  if (VEC) {
    const long long d4 = d / 4;
    for (long long j4 = tid; j4 < d4; j4 += nthreads) {
      const float4 v = *reinterpret_cast<const float4*>(row + 4 * j4);
      if (MODE == 0) {
        acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
      } else {
        const float4 u = *reinterpret_cast<const float4*>(z + 4 * j4);
        if (MODE == 1) {
          const float a = v.x - u.x, bq = v.y - u.y, c = v.z - u.z, e = v.w - u.w;
          acc += a * a + bq * bq + c * c + e * e;
        } else {
          acc += v.x * u.x + v.y * u.y + v.z * u.z + v.w * u.w;
        }
      }
    }
  }
  const long long j_begin = VEC ? (d / 4) * 4 : 0;
  for (long long j = j_begin + tid; j < d; j += nthreads) {
    const float v = row[j];
    if (MODE == 0) acc += v * v;
    else if (MODE == 1) { const float a = v - z[j]; acc += a * a; }
    else acc += v * z[j];
  }

  // wave reduce (64-wide), then LDS cross-wave, then one atomic per block
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  __shared__ float wsum[16];
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) wsum[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int i = 0; i < (int)(blockDim.x / 64); ++i) s += wsum[i];
    atomicAdd(&out[k], s);
  }
}

// ---------------------------------------------------------------------------
// K4/K8 — Gram matrix G = U U^T via f32 MFMA (v_mfma_f32_32x32x2_f32)
// ---------------------------------------------------------------------------
// grid = (tile pairs (it,jt), d-slices).  Block = 256 threads (4 waves).
// Per 32-column chunk of the slice: stage the two 32-row tiles in LDS
// (coalesced float loads, rows padded to 33 floats against bank conflicts),
// then each wave runs 4 of the 16 k-steps (2 contraction columns per MFMA)
// on its own accumulator; wave partials combine through LDS and one
// atomicAdd per output element into G (fp32 atomics: bit-level run-to-run
// variation ~1e-7 relative — documented, tested with tolerance).
// Numerics: MFMA f32 is an exact fmaf chain (guide §3), same class as a
// VALU dot product.

typedef float f32x16 __attribute__((ext_vector_type(16)));

constexpr int GRAM_BD = 32;  // contraction chunk (columns per LDS stage)

__global__ void gram_mfma_kernel(const float* __restrict__ U,
                                 float* __restrict__ G,
                                 long long K, long long d, long long ld,
                                 int ntiles, long long slice_len) {
  __shared__ float As[32][33];
  __shared__ float Bs[32][33];
  __shared__ float Gs[32][33];

  const int pair = blockIdx.x;
  const int it = pair / ntiles;
  const int jt = pair % ntiles;
  if (jt < it) return;  // symmetric: compute upper triangle, mirror on host

  const long long c_begin = (long long)blockIdx.y * slice_len;
  const long long c_end = (c_begin + slice_len < d) ? c_begin + slice_len : d;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;

  f32x16 acc = {};

  for (long long c0 = c_begin; c0 < c_end; c0 += GRAM_BD) {
    const int cols = (int)min((long long)GRAM_BD, c_end - c0);
    // stage tiles: 256 threads load 32x32 each (4 elements/thread)
    __syncthreads();
    for (int e = threadIdx.x; e < 32 * GRAM_BD; e += blockDim.x) {
      const int r = e / GRAM_BD, cc = e % GRAM_BD;
      const long long gi = (long long)it * 32 + r;
      const long long gj = (long long)jt * 32 + r;
      As[r][cc] = (gi < K && cc < cols) ? U[gi * ld + c0 + cc] : 0.f;
      Bs[r][cc] = (gj < K && cc < cols) ? U[gj * ld + c0 + cc] : 0.f;
    }
    __syncthreads();
    // 16 k-steps of 2 columns; wave w takes steps w, w+4, w+8, w+12
    const int row = lane & 31;
    const int ksel = lane >> 5;  // 0/1: which of the 2 contraction columns
    for (int s = wid; s < GRAM_BD / 2; s += 4) {
      const int c = 2 * s + ksel;
      const float a = As[row][c];
      const float b = Bs[row][c];
      acc = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc, 0, 0, 0);
    }
  }

  // combine the 4 wave partials in LDS, then one atomicAdd per element
  __syncthreads();
  for (int e = threadIdx.x; e < 32 * 33; e += blockDim.x)
    (&Gs[0][0])[e] = 0.f;
  __syncthreads();
  {
    const int col = lane & 31;
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
      atomicAdd(&Gs[row][col], acc[r]);
    }
  }
  __syncthreads();
  for (int e = threadIdx.x; e < 32 * 32; e += blockDim.x) {
    const int r = e / 32, c = e % 32;
    const long long gi = (long long)it * 32 + r;
    const long long gj = (long long)jt * 32 + c;
    if (gi < K && gj < K) {
      atomicAdd(&G[gi * K + gj], Gs[r][c]);
      if (it != jt)  // mirror off-diagonal tiles (diagonal tiles are full)
        atomicAdd(&G[gj * K + gi], Gs[r][c]);
    }
  }
}

// 64x64 tile-pair variant: halves HBM traffic (each block covers 4x the
// output area per staged chunk) and each wave OWNS one 32x32 output
// sub-tile — 4x the MFMAs per stage, no cross-wave LDS combine.
__global__ __launch_bounds__(256)
void gram_mfma64_kernel(const float* __restrict__ U,
                        float* __restrict__ G,
                        long long K, long long d, long long ld,
                        int ntiles, long long slice_len) {
  __shared__ float As[64][GRAM_BD + 1];
  __shared__ float Bs[64][GRAM_BD + 1];

  const int pair = blockIdx.x;
  const int it = pair / ntiles;
  const int jt = pair % ntiles;
  if (jt < it) return;  // symmetric: upper triangle only, mirrored below

  const long long c_begin = (long long)blockIdx.y * slice_len;
  const long long c_end = (c_begin + slice_len < d) ? c_begin + slice_len : d;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int sub_i = wid >> 1;   // wave's A-row block (0/1)
  const int sub_j = wid & 1;    // wave's B-row block (0/1)

  f32x16 acc = {};

  for (long long c0 = c_begin; c0 < c_end; c0 += GRAM_BD) {
    const int cols = (int)min((long long)GRAM_BD, c_end - c0);
    __syncthreads();
    for (int e = threadIdx.x; e < 64 * GRAM_BD; e += blockDim.x) {
      const int r = e / GRAM_BD, cc = e % GRAM_BD;
      const long long gi = (long long)it * 64 + r;
      const long long gj = (long long)jt * 64 + r;
      As[r][cc] = (gi < K && cc < cols) ? U[gi * ld + c0 + cc] : 0.f;
      Bs[r][cc] = (gj < K && cc < cols) ? U[gj * ld + c0 + cc] : 0.f;
    }
    __syncthreads();
    const int row = lane & 31;
    const int ksel = lane >> 5;  // which of the 2 contraction columns
    #pragma unroll 4
    for (int s = 0; s < GRAM_BD / 2; ++s) {
      const int c = 2 * s + ksel;
      const float a = As[32 * sub_i + row][c];
      const float b = Bs[32 * sub_j + row][c];
      acc = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc, 0, 0, 0);
    }
  }

  // each wave writes its own 32x32 sub-tile straight from the
  // accumulator (atomic across blockIdx.y slices)
  {
    const int col = lane & 31;
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int rw = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
      const long long gi = (long long)it * 64 + 32 * sub_i + rw;
      const long long gj = (long long)jt * 64 + 32 * sub_j + col;
      if (gi < K && gj < K) {
        atomicAdd(&G[gi * K + gj], acc[r]);
        if (it != jt)
          atomicAdd(&G[gj * K + gi], acc[r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

struct UView {
  const float* ptr;
  long long K, d, ld;
  bool vec;  // float4 path legal (16B-aligned rows)
};

UView view_of(const torch::Tensor& U) {
  CHECK_IN(U);
  TORCH_CHECK(U.dim() == 2, "U must be 2-D");
  TORCH_CHECK(U.stride(1) == 1, "U rows must be innermost-contiguous");
  UView v;
  v.ptr = U.data_ptr<float>();
  v.K = U.size(0);
  v.d = U.size(1);
  v.ld = U.stride(0);
  v.vec = (v.ld % 4 == 0) &&
          (reinterpret_cast<uintptr_t>(v.ptr) % 16 == 0);
  return v;
}

static inline int col_grid(long long d, int block) {
  return (int)std::min<long long>(cdiv(d, block), kMaxBlocks);
}

torch::Tensor col_mean(torch::Tensor U) {
  auto v = view_of(U);
  auto out = torch::empty({v.d}, U.options());
  const int BS = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const float invK = 1.f / (float)v.K;
  if (v.vec)
    col_reduce_kernel<0, true><<<col_grid(v.d / 4 + 1, BS), BS, 0, stream>>>(
        v.ptr, nullptr, nullptr, out.data_ptr<float>(), nullptr,
        v.K, v.d, v.ld, invK, false, (float)v.K);
  else
    col_reduce_kernel<0, false><<<col_grid(v.d, BS), BS, 0, stream>>>(
        v.ptr, nullptr, nullptr, out.data_ptr<float>(), nullptr,
        v.K, v.d, v.ld, invK, false, (float)v.K);
  return out;
}

torch::Tensor weighted_col_sum(torch::Tensor U, torch::Tensor w) {
  auto v = view_of(U);
  CHECK_IN(w);
  TORCH_CHECK(w.numel() == v.K, "weight length mismatch");
  auto wc = w.contiguous();
  auto out = torch::empty({v.d}, U.options());
  const int BS = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (v.vec)
    col_reduce_kernel<1, true><<<col_grid(v.d / 4 + 1, BS), BS, 0, stream>>>(
        v.ptr, wc.data_ptr<float>(), nullptr, out.data_ptr<float>(), nullptr,
        v.K, v.d, v.ld, 1.f, false, (float)v.K);
  else
    col_reduce_kernel<1, false><<<col_grid(v.d, BS), BS, 0, stream>>>(
        v.ptr, wc.data_ptr<float>(), nullptr, out.data_ptr<float>(), nullptr,
        v.K, v.d, v.ld, 1.f, false, (float)v.K);
  return out;
}

torch::Tensor masked_col_mean(torch::Tensor U, torch::Tensor mask,
                              double count) {
  auto v = view_of(U);
  TORCH_CHECK(mask.scalar_type() == at::kBool && mask.numel() == v.K);
  auto mc = mask.contiguous();
  // count passed by the caller: .item() here would sync the stream, which
  // is illegal under hipGraph capture (the mask is static in that regime)
  const float cnt = (float)(count >= 0 ? count : mask.sum().item<long>());
  auto out = torch::empty({v.d}, U.options());
  const int BS = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const float inv = cnt > 0 ? 1.f / cnt : 0.f;
  if (v.vec)
    col_reduce_kernel<2, true><<<col_grid(v.d / 4 + 1, BS), BS, 0, stream>>>(
        v.ptr, nullptr, mc.data_ptr<bool>(), out.data_ptr<float>(), nullptr,
        v.K, v.d, v.ld, inv, false, cnt);
  else
    col_reduce_kernel<2, false><<<col_grid(v.d, BS), BS, 0, stream>>>(
        v.ptr, nullptr, mc.data_ptr<bool>(), out.data_ptr<float>(), nullptr,
        v.K, v.d, v.ld, inv, false, cnt);
  return out;
}

std::tuple<torch::Tensor, torch::Tensor>
masked_col_mean_std(torch::Tensor U, torch::Tensor mask, bool unbiased,
                    double count) {
  auto v = view_of(U);
  TORCH_CHECK(mask.scalar_type() == at::kBool && mask.numel() == v.K);
  auto mc = mask.contiguous();
  const float cnt = (float)(count >= 0 ? count : mask.sum().item<long>());
  TORCH_CHECK(cnt >= 2, "need >=2 masked rows for std");
  auto mu = torch::empty({v.d}, U.options());
  auto sd = torch::empty({v.d}, U.options());
  const int BS = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (v.vec)
    col_reduce_kernel<3, true><<<col_grid(v.d / 4 + 1, BS), BS, 0, stream>>>(
        v.ptr, nullptr, mc.data_ptr<bool>(), mu.data_ptr<float>(),
        sd.data_ptr<float>(), v.K, v.d, v.ld, 1.f / cnt, unbiased, cnt);
  else
    col_reduce_kernel<3, false><<<col_grid(v.d, BS), BS, 0, stream>>>(
        v.ptr, nullptr, mc.data_ptr<bool>(), mu.data_ptr<float>(),
        sd.data_ptr<float>(), v.K, v.d, v.ld, 1.f / cnt, unbiased, cnt);
  return {mu, sd};
}

// General asymmetric trimmed column sum: drop b_lo smallest + b_hi
// largest per column, return inv * (sum of the rest).
static torch::Tensor trimmed_core(torch::Tensor U, long b_lo, long b_hi,
                                  double inv) {
  auto v = view_of(U);
  TORCH_CHECK(b_lo >= 0 && b_hi >= 0, "trims must be >= 0");
  TORCH_CHECK(v.K - b_lo - b_hi >= 1, "trimmed sum needs K > b_lo + b_hi");
  auto out = torch::empty({v.d}, U.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const long bmax = std::max(b_lo, b_hi);
  // small trims: divergence-free register-selection kernel (headline
  // path).  NB = smallest capacity >= bmax so a b=2 median does not pay
  // a bigger sort network.  NB=64 (2 waves/SIMD at ~210 VGPRs) covers
  // the K<=129 median that the LDS-heap kernel served divergence-bound.
  if (bmax <= 64) {
    const int BS = 256;
    const int grid = col_grid(v.d, BS);
    if (bmax <= 8)
      trimmed_regsel_kernel<8><<<grid, BS, 0, stream>>>(
          v.ptr, out.data_ptr<float>(), v.K, v.d, v.ld, (int)b_lo,
          (int)b_hi, inv);
    else if (bmax <= 16)
      trimmed_regsel_kernel<16><<<grid, BS, 0, stream>>>(
          v.ptr, out.data_ptr<float>(), v.K, v.d, v.ld, (int)b_lo,
          (int)b_hi, inv);
    else if (bmax <= 32)
      trimmed_regsel_kernel<32><<<grid, BS, 0, stream>>>(
          v.ptr, out.data_ptr<float>(), v.K, v.d, v.ld, (int)b_lo,
          (int)b_hi, inv);
    else
      trimmed_regsel_kernel<64><<<grid, BS, 0, stream>>>(
          v.ptr, out.data_ptr<float>(), v.K, v.d, v.ld, (int)b_lo,
          (int)b_hi, inv);
    return out;
  }
  // NOTE: an LDS-staged one-HBM-pass variant was measured SLOWER than
  // the 5-pass kernel at its target shapes (549 vs 207 ms at K=1000,
  // b=499, d=11.2M — 1 workgroup/CU occupancy + same-bin LDS-atomic
  // serialization outweigh the 5x HBM saving; gpurun_out/r2_call11.log).
  // It remains reachable via trimmed_mean_radix_lds for benchmarking.
  if (false && v.K <= RT_L_KMAX) {
    const size_t smem = (size_t)RT_L * 8 + 6 * RT_L * 4
        + 2 * (size_t)RT_L * 256 * 4 + (size_t)v.K * RT_L * 4;
    static bool attr_set = false;
    if (!attr_set) {
      hipFuncSetAttribute(
          reinterpret_cast<const void*>(&radix_trimmed_lds_kernel),
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
      attr_set = true;
    }
    const int grid = (int)std::min<long long>(cdiv(v.d, RT_L), kMaxBlocks);
    radix_trimmed_lds_kernel<<<grid, 256, smem, stream>>>(
        v.ptr, out.data_ptr<float>(), v.K, v.d, v.ld, b_lo, b_hi, inv);
    return out;
  }
  // mid symmetric b: LDS-heap streaming selection (2*b per-thread LDS)
  if (b_lo == b_hi && inv != 1.0) {
    int BS = 256;
    size_t need = 2 * (size_t)b_lo * 4 * BS;
    while (BS > 64 && need > 64 * 1024) { BS /= 2; need /= 2; }
    if (need <= 64 * 1024) {
      const int grid = col_grid(v.d, BS);
      trimmed_select_kernel<<<grid, BS, need, stream>>>(
          v.ptr, out.data_ptr<float>(), v.K, v.d, v.ld, (int)b_lo);
      return out;
    }
  }
  // large / asymmetric trims: dual radix-select kernel
  const int grid = (int)std::min<long long>(cdiv(v.d, RT_T), kMaxBlocks);
  radix_trimmed_kernel<<<grid, 256, 0, stream>>>(
      v.ptr, out.data_ptr<float>(), v.K, v.d, v.ld, b_lo, b_hi, inv);
  return out;
}

torch::Tensor trimmed_mean_select(torch::Tensor U, long b) {
  TORCH_CHECK(U.size(0) - 2 * b >= 1, "trimmed_mean needs K > 2b");
  TORCH_CHECK(b >= 0, "b must be >= 0");
  if (b == 0) {
    return col_mean(U);
  }
  return trimmed_core(U, b, b, 1.0 / (double)(U.size(0) - 2 * b));
}

torch::Tensor col_trimmed_sum(torch::Tensor U, long b_lo, long b_hi) {
  return trimmed_core(U, b_lo, b_hi, 1.0);
}

torch::Tensor trimmed_mean(torch::Tensor U, long b) {
  return trimmed_mean_select(U, b);
}

torch::Tensor trimmed_mean_radix_lds(torch::Tensor U, long b) {
  // benchmark-only entry for the LDS-staged radix (see note in
  // trimmed_core: measured slower than the 5-pass kernel)
  auto v = view_of(U);
  TORCH_CHECK(v.K - 2 * b >= 1 && b > 0 && v.K <= RT_L_KMAX, "bad shape");
  auto out = torch::empty({v.d}, U.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const size_t smem = (size_t)RT_L * 8 + 6 * RT_L * 4
      + 2 * (size_t)RT_L * 256 * 4 + (size_t)v.K * RT_L * 4;
  hipFuncSetAttribute(
      reinterpret_cast<const void*>(&radix_trimmed_lds_kernel),
      hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
  const int grid = (int)std::min<long long>(cdiv(v.d, RT_L), kMaxBlocks);
  radix_trimmed_lds_kernel<<<grid, 256, smem, stream>>>(
      v.ptr, out.data_ptr<float>(), v.K, v.d, v.ld, b, b,
      1.0 / (double)(v.K - 2 * b));
  return out;
}

torch::Tensor trimmed_mean_radix(torch::Tensor U, long b) {
  // direct radix path (A/B benchmarking; auto-dispatch uses the register
  // selection kernel for small b)
  auto v = view_of(U);
  TORCH_CHECK(v.K - 2 * b >= 1 && b > 0, "bad b");
  auto out = torch::empty({v.d}, U.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int grid = (int)std::min<long long>(cdiv(v.d, RT_T), kMaxBlocks);
  radix_trimmed_kernel<<<grid, 256, 0, stream>>>(
      v.ptr, out.data_ptr<float>(), v.K, v.d, v.ld, b, b,
      1.0 / (double)(v.K - 2 * b));
  return out;
}

torch::Tensor col_median(torch::Tensor U) {
  auto v = view_of(U);
  const long b = (v.K - 1) / 2;  // K-2b = 1 (odd K) or 2 (even K: avg both)
  return trimmed_mean_select(U, b);
}

static torch::Tensor row_reduce(torch::Tensor U, const float* z, int mode) {
  auto v = view_of(U);
  auto out = torch::zeros({v.K}, U.options());
  const int BS = 256;
  // enough blocks to fill the chip even at small K
  const int splits = (int)std::max<long long>(
      1, std::min<long long>(cdiv(v.d, 4 * BS), cdiv(2048, v.K)));
  dim3 grid(splits, (unsigned)v.K);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (v.vec) {
    if (mode == 0)
      row_reduce_kernel<0, true><<<grid, BS, 0, stream>>>(v.ptr, z, out.data_ptr<float>(), v.K, v.d, v.ld);
    else if (mode == 1)
      row_reduce_kernel<1, true><<<grid, BS, 0, stream>>>(v.ptr, z, out.data_ptr<float>(), v.K, v.d, v.ld);
    else
      row_reduce_kernel<2, true><<<grid, BS, 0, stream>>>(v.ptr, z, out.data_ptr<float>(), v.K, v.d, v.ld);
  } else {
    if (mode == 0)
      row_reduce_kernel<0, false><<<grid, BS, 0, stream>>>(v.ptr, z, out.data_ptr<float>(), v.K, v.d, v.ld);
    else if (mode == 1)
      row_reduce_kernel<1, false><<<grid, BS, 0, stream>>>(v.ptr, z, out.data_ptr<float>(), v.K, v.d, v.ld);
    else
      row_reduce_kernel<2, false><<<grid, BS, 0, stream>>>(v.ptr, z, out.data_ptr<float>(), v.K, v.d, v.ld);
  }
  return out;
}

torch::Tensor row_sq_norms(torch::Tensor U) {
  return row_reduce(U, nullptr, 0);
}

torch::Tensor row_diff_sq_norms(torch::Tensor U, torch::Tensor z) {
  CHECK_IN(z);
  auto zc = z.contiguous();
  TORCH_CHECK(zc.numel() == U.size(1), "z length mismatch");
  return row_reduce(U, zc.data_ptr<float>(), 1);
}

torch::Tensor row_dots(torch::Tensor U, torch::Tensor vv) {
  CHECK_IN(vv);
  auto vc = vv.contiguous();
  TORCH_CHECK(vc.numel() == U.size(1), "v length mismatch");
  return row_reduce(U, vc.data_ptr<float>(), 2);
}

static torch::Tensor gram_t32(torch::Tensor U) {
  auto v = view_of(U);
  auto G = torch::zeros({v.K, v.K}, U.options());
  const int ntiles = (int)cdiv(v.K, 32);
  const int npairs = ntiles * ntiles;  // lower triangle returns early
  // slice d so total blocks ~ >=1024 for occupancy
  long long nsl = std::max<long long>(1, 1024 / std::max(1, npairs));
  long long slice = std::max<long long>(GRAM_BD, cdiv(v.d, nsl));
  slice = cdiv(slice, GRAM_BD) * GRAM_BD;
  nsl = cdiv(v.d, slice);
  dim3 grid((unsigned)npairs, (unsigned)nsl);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  gram_mfma_kernel<<<grid, 256, 0, stream>>>(
      v.ptr, G.data_ptr<float>(), v.K, v.d, v.ld, ntiles, slice);
  return G;
}

static torch::Tensor gram_t64(torch::Tensor U) {
  auto v = view_of(U);
  auto G = torch::zeros({v.K, v.K}, U.options());
  const int ntiles = (int)cdiv(v.K, 64);
  const int npairs = ntiles * ntiles;
  long long nsl = std::max<long long>(1, 1024 / std::max(1, npairs));
  long long slice = std::max<long long>(GRAM_BD, cdiv(v.d, nsl));
  slice = cdiv(slice, GRAM_BD) * GRAM_BD;
  nsl = cdiv(v.d, slice);
  dim3 grid((unsigned)npairs, (unsigned)nsl);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  gram_mfma64_kernel<<<grid, 256, 0, stream>>>(
      v.ptr, G.data_ptr<float>(), v.K, v.d, v.ld, ntiles, slice);
  return G;
}

torch::Tensor gram(torch::Tensor U) {
  // 64x64 tile pairs halve the HBM traffic (K²d/64 vs K²d/32 reads) and
  // give each wave 4x the MFMAs per LDS stage; below ~3 tiles the small
  // kernel fills the chip better.
  return (U.size(0) >= 96) ? gram_t64(U) : gram_t32(U);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "blades_amd CDNA4 HIP kernels (gfx950)";
  m.def("col_mean", &col_mean, "column mean (K1)");
  m.def("weighted_col_sum", &weighted_col_sum, "weighted column sum");
  m.def("masked_col_mean", &masked_col_mean, "masked column mean (K11)",
        py::arg("U"), py::arg("mask"), py::arg("count") = -1.0);
  m.def("masked_col_mean_std", &masked_col_mean_std,
        "masked column mean+std (K10)", py::arg("U"), py::arg("mask"),
        py::arg("unbiased") = true, py::arg("count") = -1.0);
  m.def("trimmed_mean", &trimmed_mean, "coordinate-wise trimmed mean (K3)");
  m.def("col_trimmed_sum", &col_trimmed_sum,
        "asymmetric trimmed column sum (K5 krum scores)");
  m.def("trimmed_mean_radix_lds", &trimmed_mean_radix_lds,
        "LDS-staged radix (benchmark-only; slower, kept for evidence)");
  m.def("trimmed_mean_radix", &trimmed_mean_radix,
        "trimmed mean via dual radix select (benchmarking entry)");
  m.def("col_median", &col_median, "coordinate-wise median (K2)");
  m.def("row_sq_norms", &row_sq_norms, "per-row squared norms");
  m.def("row_diff_sq_norms", &row_diff_sq_norms,
        "per-row squared distance to z (K6)");
  m.def("row_dots", &row_dots, "per-row dot with v (K9)");
  m.def("gram", &gram, "U U^T via f32 MFMA (K4/K8)");
}
