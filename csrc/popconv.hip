// Population (many-model) 3x3 convolution — CDNA4 MFMA f32 direct kernels.
//
// The framework trains the whole simulated client population as one batched
// computation.  MIOpen has no batched-weight convolution: the vmapped path
// decomposes into ~1e5 per-client im2col+GEMM kernels per round
// (profiles/r01_vmap_resnet18_kernel_stats.csv).  These kernels implement
// the population conv DIRECTLY on the matrix cores:
//
//   activations: [C, ch, Np] fp32, Np = B*Hp*Wp with 1-pixel zero-padded
//                planes (Hp=H+2, Wp=W+2).  Zero pads make every 3x3 tap a
//                pure flat pointer shift Δ = dy*Wp + dx: output position
//                n = (b, y, x) reads n+Δ within the same padded plane for
//                all interior n, so conv = 9 shifted GEMMs with NO im2col.
//   weights:     [C, co, ci, 3, 3] per client, or stride-0 broadcast over
//                C when all clients share θ (FedSGD).
//
// popconv_fwd:  Y[c,co,n] = Σ_t Σ_ci W[c,co,ci,t] · X[c,ci,n+Δt]
//   grid (n_tiles, co_tiles, C); block 256 (4 waves), wave = one 32co×32n
//   MFMA f32 (v_mfma_f32_32x32x2_f32) accumulator; X staged per 32-ci chunk
//   in LDS with halo (row length NT+2*Wp+2), W per (chunk, tap) in LDS with
//   +1 row padding against bank conflicts.  All n are computed (pad columns
//   accumulate cross-image garbage) and the caller zeroes pad columns — the
//   interior result is exact.  Also used for dX with transposed/flipped
//   weights prepared host-side (dX = conv(dY, Wᵀ flipped), dY pads zeroed).
//
// popconv_dw:  dW[c,co,ci,t] = Σ_n dY[c,co,n] · X[c,ci,n+Δt]
//   grid (co_t×ci_t, C*9, kslices); block 256 (4 waves k-interleaved),
//   LDS-staged 32×64 dY/X tiles, wave partials combined in LDS then one
//   atomicAdd per element (fp32 atomics: ~1e-7 run-to-run jitter,
//   documented).  dY pad columns MUST be zeroed by the caller first (pad
//   outputs are constants, their upstream grads are meaningless).
//
// Numerics: MFMA f32 is an exact fmaf chain (guide §3) — same class as any
// fp32 conv; parity tests vs torch.nn.grad live in tests/test_popconv.py.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstdint>

namespace {

typedef float f32x16 __attribute__((ext_vector_type(16)));

constexpr int NT = 256;        // n-tile per block (4 waves × 2×32)
constexpr int CK = 32;         // ci chunk
constexpr int MAX_WP = 40;     // max padded width (CIFAR: 34)
constexpr int XROW = NT + 2 * MAX_WP + 2 + 2;  // LDS X row length (+2 bank skew)

#define CHECK_F32(x)                                                   \
  TORCH_CHECK((x).is_cuda() && (x).scalar_type() == at::kFloat,        \
              #x " must be fp32 GPU tensor");

__global__ __launch_bounds__(256)
void popconv_fwd_kernel(const float* __restrict__ X,
                        const float* __restrict__ W,
                        float* __restrict__ Y,
                        int C, int co, int ci, long long Np, int Wp,
                        long long strideWc /* 0 when shared */) {
  // LDS: X chunk with halo (CK x XROW ≈ 42 KiB) + all nine taps' weights
  // for the chunk (9 x 32 x 33 ≈ 38 KiB) -> 2 barriers per ci chunk
  // instead of 2 per (chunk, tap).
  __shared__ float Xs[CK][XROW];
  __shared__ float Ws[9][32][33];

  const int c = blockIdx.z;
  const int co0 = blockIdx.y * 32;
  const long long n0 = (long long)blockIdx.x * NT;
  const int halo = Wp + 1;

  const float* Xc = X + (long long)c * ci * Np;
  const float* Wc = W + (long long)c * strideWc;
  float* Yc = Y + (long long)c * co * Np;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int arow = lane & 31;       // A row (co) / B col (n)
  const int ksel = lane >> 5;       // which of the 2 contraction cols

  f32x16 acc0 = {};                 // wave's n sub-tiles: wid*64 + {0,32}
  f32x16 acc1 = {};

  const int n_chunks = (ci + CK - 1) / CK;
  for (int cc = 0; cc < n_chunks; ++cc) {
    const int ci0 = cc * CK;
    __syncthreads();
    {
      const int cols = NT + 2 * halo;
      for (int e = threadIdx.x; e < CK * cols; e += blockDim.x) {
        const int r = e / cols, q = e % cols;
        const long long n = n0 - halo + q;
        const int cii = ci0 + r;
        Xs[r][q] = (cii < ci && n >= 0 && n < Np)
                       ? Xc[(long long)cii * Np + n] : 0.f;
      }
      for (int e = threadIdx.x; e < 9 * 32 * CK; e += blockDim.x) {
        const int t = e / (32 * CK);
        const int r = (e / CK) % 32, k = e % CK;
        const int coo = co0 + r, cii = ci0 + k;
        Ws[t][r][k] = (coo < co && cii < ci)
                          ? Wc[(((long long)coo * ci + cii) * 9) + t] : 0.f;
      }
    }
    __syncthreads();
    #pragma unroll
    for (int t = 0; t < 9; ++t) {
      const int dy = t / 3 - 1, dx = t % 3 - 1;
      const int delta = dy * Wp + dx;
      const int nn = halo + delta + wid * 64 + arow;
      #pragma unroll
      for (int k2 = 0; k2 < CK / 2; ++k2) {
        const float a = Ws[t][arow][2 * k2 + ksel];
        const float b0 = Xs[2 * k2 + ksel][nn];
        const float b1 = Xs[2 * k2 + ksel][nn + 32];
        acc0 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b0, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b1, acc1, 0, 0, 0);
      }
    }
  }

  // ---- epilogue: D[r] -> row (r&3)+8*(r>>2)+4*(lane>>5), col lane&31
  #pragma unroll
  for (int half = 0; half < 2; ++half) {
    const long long nbase = n0 + wid * 64 + half * 32 + (lane & 31);
    if (nbase >= Np) continue;
    const f32x16& acc = half ? acc1 : acc0;
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
      const int coo = co0 + row;
      if (coo < co) Yc[(long long)coo * Np + nbase] = acc[r];
    }
  }
}

__global__ __launch_bounds__(256)
void popconv_dw_kernel(const float* __restrict__ dY,
                          const float* __restrict__ X,
                          float* __restrict__ dW,
                          int C, int co, int ci, long long Np, int Wp,
                          long long kslice_len, long long strideWc) {
  __shared__ float Ys[32][66];
  __shared__ float Xs[32][66];
  __shared__ float Gs[32][33];

  const int ci_tiles = (ci + 31) / 32;
  const int co0 = (blockIdx.x / ci_tiles) * 32;
  const int ci0 = (blockIdx.x % ci_tiles) * 32;
  const int c = blockIdx.y / 9;
  const int t = blockIdx.y % 9;
  const int dy_ = t / 3 - 1, dx_ = t % 3 - 1;
  const int delta = dy_ * Wp + dx_;
  const long long k_begin = (long long)blockIdx.z * kslice_len;
  const long long k_end = (k_begin + kslice_len < Np) ? k_begin + kslice_len : Np;

  const float* Yc = dY + (long long)c * co * Np;
  const float* Xc = X + (long long)c * ci * Np;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int arow = lane & 31;
  const int ksel = lane >> 5;

  f32x16 acc = {};

  for (long long k0 = k_begin; k0 < k_end; k0 += 64) {
    __syncthreads();
    for (int e = threadIdx.x; e < 32 * 64; e += blockDim.x) {
      const int r = e / 64, q = e % 64;
      const long long k = k0 + q;
      const int coo = co0 + r;
      Ys[r][q] = (coo < co && k < Np) ? Yc[(long long)coo * Np + k] : 0.f;
      const int cii = ci0 + r;
      const long long kx = k + delta;
      Xs[r][q] = (cii < ci && k < Np && kx >= 0 && kx < Np)
                     ? Xc[(long long)cii * Np + kx] : 0.f;
    }
    __syncthreads();
    // 64-wide contraction split over 4 waves: wave w takes k = 16w .. 16w+15
    const int kb = wid * 16;
    #pragma unroll
    for (int k2 = 0; k2 < 8; ++k2) {
      const float a = Ys[arow][kb + 2 * k2 + ksel];
      const float b = Xs[arow][kb + 2 * k2 + ksel];
      // A[co][k] = Ys, B[k][ci] = Xs^T: lane's B col (arow) indexes ci
      acc = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc, 0, 0, 0);
    }
  }

  // combine wave partials in LDS, then one atomicAdd per element
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
  float* dWc = dW + (long long)c * strideWc;
  for (int e = threadIdx.x; e < 32 * 32; e += blockDim.x) {
    const int r = e / 32, q = e % 32;
    const int coo = co0 + r, cii = ci0 + q;
    if (coo < co && cii < ci)
      atomicAdd(&dWc[(((long long)coo * ci + cii) * 9) + t], Gs[r][q]);
  }
}

// ------------------------------------------------------------------ host

torch::Tensor popconv_fwd(torch::Tensor X, torch::Tensor W, long B,
                          long Hp, long Wp) {
  CHECK_F32(X);
  CHECK_F32(W);
  TORCH_CHECK(X.dim() == 3, "X must be [C, ci, Np]");
  TORCH_CHECK(W.dim() == 5, "W must be [C, co, ci, 3, 3]");
  TORCH_CHECK(W.size(3) == 3 && W.size(4) == 3, "3x3 kernels only");
  TORCH_CHECK(X.stride(2) == 1 && X.stride(1) == X.size(2),
              "X rows must be contiguous");
  TORCH_CHECK(Wp + 1 <= MAX_WP, "padded width too large for LDS halo");
  const long long C = X.size(0), ci = X.size(1), Np = X.size(2);
  const long long co = W.size(1);
  TORCH_CHECK(W.size(2) == ci, "ci mismatch");
  TORCH_CHECK(Np == B * Hp * Wp, "Np mismatch");
  // W's client dim may be: stride-0 (shared θ expand), contiguous, or the
  // flat-slab row stride d (ParamSpec batched views) — the kernel only
  // needs the per-client inner layout contiguous
  long long strideWc = W.stride(0);
  TORCH_CHECK(strideWc == 0 || strideWc >= co * ci * 9,
              "W client stride must be 0 (broadcast) or >= co*ci*9");
  TORCH_CHECK(W.stride(1) == ci * 9 && W.stride(4) == 1,
              "W inner layout must be contiguous");

  auto Y = torch::empty({(long)C, (long)co, (long)Np}, X.options());
  dim3 grid((unsigned)((Np + NT - 1) / NT), (unsigned)((co + 31) / 32),
            (unsigned)C);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  popconv_fwd_kernel<<<grid, 256, 0, stream>>>(
      X.data_ptr<float>(), W.data_ptr<float>(), Y.data_ptr<float>(),
      (int)C, (int)co, (int)ci, Np, (int)Wp, strideWc);
  return Y;
}

torch::Tensor popconv_dw(torch::Tensor dY, torch::Tensor X, long B,
                         long Hp, long Wp, bool shared) {
  CHECK_F32(dY);
  CHECK_F32(X);
  const long long C = X.size(0), ci = X.size(1), Np = X.size(2);
  const long long co = dY.size(1);
  TORCH_CHECK(dY.size(0) == C && dY.size(2) == Np, "shape mismatch");
  TORCH_CHECK(dY.stride(2) == 1 && dY.stride(1) == Np, "dY rows contiguous");
  TORCH_CHECK(X.stride(2) == 1 && X.stride(1) == Np, "X rows contiguous");

  auto dW = torch::zeros({(long)C, (long)co, (long)ci, 3, 3}, X.options());
  const int co_t = (int)((co + 31) / 32), ci_t = (int)((ci + 31) / 32);
  // pick kslices so total blocks ≈ >=1536
  long long want = 1536 / std::max<long long>(1, (long long)co_t * ci_t * C * 9);
  long long kslices = std::max<long long>(1, want);
  long long kslice_len = ((Np + kslices - 1) / kslices + 63) / 64 * 64;
  kslices = (Np + kslice_len - 1) / kslice_len;
  dim3 grid((unsigned)(co_t * ci_t), (unsigned)(C * 9), (unsigned)kslices);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  popconv_dw_kernel<<<grid, 256, 0, stream>>>(
      dY.data_ptr<float>(), X.data_ptr<float>(), dW.data_ptr<float>(),
      (int)C, (int)co, (int)ci, Np, (int)Wp, kslice_len, co * ci * 9);
  return dW;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "blades_amd population-conv MFMA kernels (gfx950)";
  m.def("popconv_fwd", &popconv_fwd,
        "population 3x3 conv forward / dX (shifted-tap MFMA f32)");
  m.def("popconv_dw", &popconv_dw,
        "population 3x3 conv per-client weight grad (MFMA f32 split-K)");
}
