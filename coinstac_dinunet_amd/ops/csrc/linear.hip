// K2 — MFMA-tiled linear with fused bias(+ReLU) epilogue.
// The MLP family's hidden layers run linear -> bias -> ReLU as one kernel
// (3 launches + 2 extra HBM round trips in stock eager). GEMM-shaped work
// belongs on the matrix cores: this uses v_mfma_f32_16x16x4_f32 (exact
// f32 at the 157 TF vector-peak rate; fragment layout per
// cdna_hip_programming.md §3: A[l&15][l>>4], B[l>>4][l&15],
// C col=l&15, row=(l>>4)*4+reg).
//
// Geometry: 256-thread block = 4 waves as 2x2, each wave one 32x32 C tile
// via 2x2 mfma fragments => 64x64 block tile; A/B staged through LDS with
// +1-float row padding (no bank conflicts); K stepped by 16.
// Inputs may be fp32 or bf16 (upconverted on load); accumulate is f32.
#include "common.h"

#include <hip/hip_bf16.h>

using f32x4 = __attribute__((ext_vector_type(4))) float;

#define BM 64
#define BN 64
#define BK 16
#define LDK (BK + 1)  // +1 pad: column reads hit distinct banks

template <typename T>
__device__ inline float ldf(const T* p) { return (float)*p; }
template <>
__device__ inline float ldf<__hip_bfloat16>(const __hip_bfloat16* p) {
  return __bfloat162float(*p);
}

template <typename T>
__device__ inline T stf(float v) { return (T)v; }
template <>
__device__ inline __hip_bfloat16 stf<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// C[M,N] = A[M,K] (row-major, lda) @ B[N,K]^T (row-major, ldb = weight
// layout) + bias; optional ReLU. One wave computes 32x32 via 2x2 fragments.
// TC = output element type (bf16 epilogue writes directly — no fp32
// round-trip tensor).
template <typename TA, typename TB, typename TC, bool RELU, bool HAS_BIAS>
__global__ __launch_bounds__(256) void linear_fwd_kernel(
    const TA* __restrict__ A, const TB* __restrict__ B,
    const float* __restrict__ bias, TC* __restrict__ C, int M, int N,
    int K, int lda, int ldb, int ldc) {
  __shared__ float sA[BM][LDK];
  __shared__ float sB[BN][LDK];

  const int bm = blockIdx.x * BM;
  const int bn = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;   // 0..3
  const int lane = tid % WAVE_SIZE;
  const int wm = (wave >> 1) * 32;    // wave row offset in block tile
  const int wn = (wave & 1) * 32;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += BK) {
    // stage A[BM][BK] and B[BN][BK]: 256 threads x 4 elements each
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int idx = tid + r * 256;          // 0..1023 = BM*BK
      int row = idx / BK, col = idx % BK;
      int gm = bm + row, gk = k0 + col;
      sA[row][col] = (gm < M && gk < K) ? ldf(&A[(int64_t)gm * lda + gk]) : 0.f;
      int gn = bn + row;
      sB[row][col] = (gn < N && gk < K) ? ldf(&B[(int64_t)gn * ldb + gk]) : 0.f;
    }
    __syncthreads();

    // mfma_f32_16x16x4: lane l feeds A[l&15][l>>4], B[l>>4][l&15]
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const int ar = lane & 15, ak = lane >> 4;  // ak in 0..3
#pragma unroll
      for (int i = 0; i < 2; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          float a = sA[wm + i * 16 + ar][kk + ak];
          float b = sB[wn + j * 16 + ar][kk + ak];  // B^T: row=n, col=k
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[i][j],
                                                           0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // epilogue: C row=(l>>4)*4+reg, col=l&15
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int gm = bm + wm + i * 16 + crow0 + r;
        int gn = bn + wn + j * 16 + ccol;
        if (gm < M && gn < N) {
          float v = acc[i][j][r];
          if (HAS_BIAS) v += bias[gn];
          if (RELU) v = fmaxf(v, 0.f);
          C[(int64_t)gm * ldc + gn] = stf<TC>(v);
        }
      }
    }
  }
}

template <typename TA, typename TB, typename TC>
static void launch_linear(const TA* A, const TB* B, const float* bias,
                          TC* C, int M, int N, int K, int lda, int ldb,
                          int ldc, bool relu, hipStream_t stream) {
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  dim3 block(256);
  if (bias) {
    if (relu)
      hipLaunchKernelGGL((linear_fwd_kernel<TA, TB, TC, true, true>), grid,
                         block, 0, stream, A, B, bias, C, M, N, K, lda, ldb,
                         ldc);
    else
      hipLaunchKernelGGL((linear_fwd_kernel<TA, TB, TC, false, true>), grid,
                         block, 0, stream, A, B, bias, C, M, N, K, lda, ldb,
                         ldc);
  } else {
    if (relu)
      hipLaunchKernelGGL((linear_fwd_kernel<TA, TB, TC, true, false>), grid,
                         block, 0, stream, A, B, nullptr, C, M, N, K, lda,
                         ldb, ldc);
    else
      hipLaunchKernelGGL((linear_fwd_kernel<TA, TB, TC, false, false>), grid,
                         block, 0, stream, A, B, nullptr, C, M, N, K, lda,
                         ldb, ldc);
  }
}

torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor weight,
                         torch::Tensor bias, bool relu) {
  CHECK_GPU(x);
  auto x2 = x.contiguous();
  auto w = weight.contiguous();
  bool has_bias = bias.numel() > 0;
  auto b = has_bias ? bias.to(torch::kFloat32).contiguous() : bias;
  int64_t M = x2.size(0), K = x2.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "weight/input K mismatch");
  auto out = torch::empty({M, N}, x2.options());
  auto stream = current_stream();
  const float* bp = has_bias ? b.data_ptr<float>() : nullptr;

  if (x2.scalar_type() == torch::kFloat32 &&
      w.scalar_type() == torch::kFloat32) {
    launch_linear(x2.data_ptr<float>(), w.data_ptr<float>(), bp,
                  out.data_ptr<float>(), (int)M, (int)N, (int)K, (int)K,
                  (int)K, (int)N, relu, stream);
  } else if (x2.scalar_type() == torch::kBFloat16 &&
             w.scalar_type() == torch::kBFloat16) {
    launch_linear(reinterpret_cast<const __hip_bfloat16*>(x2.data_ptr()),
                  reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()), bp,
                  reinterpret_cast<__hip_bfloat16*>(out.data_ptr()), (int)M,
                  (int)N, (int)K, (int)K, (int)K, (int)N, relu, stream);
  } else {
    TORCH_CHECK(false, "linear_fwd: unsupported dtype combination");
  }
  return out;
}

// ---------------------------------------------------------------------------
// K2 backward — MFMA kernels for the two backward GEMMs (replaces the
// rocBLAS calls the round-1 build used).
//   dgrad:  gx[M,K] = go[M,N] @ W[N,K]        (contraction over N, A direct)
//   wgrad:  gw[N,K] = go[M,N]^T @ x[M,K]      (contraction over M, A transp)
// Unified form: C[i,j] = sum_c opA(i,c) * B[c,j] with B row-major [C,ldb]
// contracted along its rows. Staging differs from the fwd kernel: B tiles
// are read coalesced along j and written transposed into LDS.
// ---------------------------------------------------------------------------
template <typename TA, typename TB, typename TC, bool TRANS_A,
          bool SPLIT_K = false>
__global__ __launch_bounds__(256) void linear_bwd_kernel(
    const TA* __restrict__ A, const TB* __restrict__ B, TC* __restrict__ C,
    int M, int N, int K, int lda, int ldb, int ldc, int kchunk = 0) {
  // M = rows of C, N = cols of C, K = contraction length. SPLIT_K:
  // grid.z splits the contraction (skinny GEMMs whose big dimension is
  // K — e.g. MLP wgrad [256,66] over batch 1024 — otherwise collapse to
  // a handful of blocks); partials fold by fp32 atomics, C pre-zeroed.
  __shared__ float sA[BM][LDK];
  __shared__ float sB[BN][LDK];

  const int bm = blockIdx.x * BM;
  const int bn = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int wm = (wave >> 1) * 32;
  const int wn = (wave & 1) * 32;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  int kbeg = 0, kend = K;
  if (SPLIT_K) {
    kbeg = blockIdx.z * kchunk;
    kend = min(K, kbeg + kchunk);
  }
  for (int k0 = kbeg; k0 < kend; k0 += BK) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int idx = tid + r * 256;  // 0..1023 = BM*BK
      if (TRANS_A) {
        // A[c,i] layout: read coalesced along i, store sA[i][c]
        int c = idx / BM, row = idx % BM;
        int gm = bm + row, gk = k0 + c;
        sA[row][c] =
            (gm < M && gk < K) ? ldf(&A[(int64_t)gk * lda + gm]) : 0.f;
      } else {
        int row = idx / BK, col = idx % BK;
        int gm = bm + row, gk = k0 + col;
        sA[row][col] =
            (gm < M && gk < K) ? ldf(&A[(int64_t)gm * lda + gk]) : 0.f;
      }
      // B[c,j]: read coalesced along j, store transposed sB[j][c]
      int c = idx / BN, col = idx % BN;
      int gn = bn + col, gk = k0 + c;
      sB[col][c] = (gn < N && gk < K) ? ldf(&B[(int64_t)gk * ldb + gn]) : 0.f;
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const int ar = lane & 15, ak = lane >> 4;
#pragma unroll
      for (int i = 0; i < 2; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          float a = sA[wm + i * 16 + ar][kk + ak];
          float b = sB[wn + j * 16 + ar][kk + ak];
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[i][j],
                                                           0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int gm = bm + wm + i * 16 + crow0 + r;
        int gn = bn + wn + j * 16 + ccol;
        if (gm < M && gn < N) {
          if (SPLIT_K)
            atomicAdd(reinterpret_cast<float*>(&C[(int64_t)gm * ldc + gn]),
                      acc[i][j][r]);
          else
            C[(int64_t)gm * ldc + gn] = stf<TC>(acc[i][j][r]);
        }
      }
}

template <typename TA, typename TB, typename TC, bool TRANS_A>
static void launch_linear_bwd(const TA* A, const TB* B, TC* C, int M, int N,
                              int K, int lda, int ldb, int ldc,
                              hipStream_t stream) {
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  hipLaunchKernelGGL((linear_bwd_kernel<TA, TB, TC, TRANS_A>), grid,
                     dim3(256), 0, stream, A, B, C, M, N, K, lda, ldb, ldc);
}

// gx[M,K] = go[M,N] @ W[N,K]; output dtype follows go.
torch::Tensor linear_dgrad(torch::Tensor go, torch::Tensor weight) {
  CHECK_GPU(go);
  auto g = go.contiguous();
  auto w = weight.to(g.scalar_type()).contiguous();
  int64_t M = g.size(0), N = g.size(1), K = w.size(1);
  TORCH_CHECK(w.size(0) == N, "weight/grad N mismatch");
  auto gx = torch::empty({M, K}, g.options());
  auto stream = current_stream();
  if (g.scalar_type() == torch::kFloat32) {
    launch_linear_bwd<float, float, float, false>(
        g.data_ptr<float>(), w.data_ptr<float>(), gx.data_ptr<float>(),
        (int)M, (int)K, (int)N, (int)N, (int)K, (int)K, stream);
  } else if (g.scalar_type() == torch::kBFloat16) {
    launch_linear_bwd<__hip_bfloat16, __hip_bfloat16, __hip_bfloat16, false>(
        reinterpret_cast<const __hip_bfloat16*>(g.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
        reinterpret_cast<__hip_bfloat16*>(gx.data_ptr()), (int)M, (int)K,
        (int)N, (int)N, (int)K, (int)K, stream);
  } else {
    TORCH_CHECK(false, "linear_dgrad: fp32 or bf16 only");
  }
  return gx;
}

// gw[N,K] = go[M,N]^T @ x[M,K]; fp32 out (master-weight gradient).
// Skinny output over a long contraction (the MLP shape) uses split-K.
torch::Tensor linear_wgrad(torch::Tensor go, torch::Tensor x) {
  CHECK_GPU(go);
  auto g = go.contiguous();
  auto x2 = x.to(g.scalar_type()).contiguous();
  int64_t M = g.size(0), N = g.size(1), K = x2.size(1);
  TORCH_CHECK(x2.size(0) == M, "x/grad M mismatch");
  auto stream = current_stream();
  int64_t tiles = ((N + BM - 1) / BM) * ((K + BN - 1) / BN);
  int nsplit = 1;
  if (tiles < 256 && M > 4 * BK) {
    nsplit = (int)std::min<int64_t>((M + BK - 1) / BK,
                                    std::max<int64_t>(1, 512 / tiles));
  }
  const bool split = nsplit > 1;
  auto gw = split
      ? torch::zeros({N, K}, g.options().dtype(torch::kFloat32))
      : torch::empty({N, K}, g.options().dtype(torch::kFloat32));
  int kchunk = (int)(((M + (int64_t)nsplit * BK - 1) /
                      ((int64_t)nsplit * BK)) * BK);
  dim3 grid((unsigned)((N + BM - 1) / BM), (unsigned)((K + BN - 1) / BN),
            (unsigned)((M + kchunk - 1) / kchunk));
  auto L = [&](auto AB, auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, stream, AB.first,
                       AB.second, gw.data_ptr<float>(), (int)N, (int)K,
                       (int)M, (int)N, (int)K, (int)K, kchunk);
  };
  if (g.scalar_type() == torch::kFloat32) {
    auto ab = std::make_pair(g.data_ptr<float>(), x2.data_ptr<float>());
    if (split) L(ab, (linear_bwd_kernel<float, float, float, true, true>));
    else L(ab, (linear_bwd_kernel<float, float, float, true, false>));
  } else if (g.scalar_type() == torch::kBFloat16) {
    auto ab = std::make_pair(
        reinterpret_cast<const __hip_bfloat16*>(g.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(x2.data_ptr()));
    if (split)
      L(ab, (linear_bwd_kernel<__hip_bfloat16, __hip_bfloat16, float, true,
                               true>));
    else
      L(ab, (linear_bwd_kernel<__hip_bfloat16, __hip_bfloat16, float, true,
                               false>));
  } else {
    TORCH_CHECK(false, "linear_wgrad: fp32 or bf16 only");
  }
  return gw;
}

// bias gradient: column sum of grad_out [M,N] -> [N]. One 256-thread
// block per column, threads stride the rows, wave+LDS reduce (the old
// one-thread-per-column loop serialized the whole tensor into a single
// block at MLP shapes — 144 us for [1024, 256]).
__global__ void colsum_kernel(const float* __restrict__ g,
                              float* __restrict__ out, int64_t M, int64_t N) {
  const int64_t n = blockIdx.x;
  float s = 0.f;
  for (int64_t m = threadIdx.x; m < M; m += blockDim.x)
    s += g[m * N + n];
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
    s += __shfl_down(s, off);
  __shared__ float part[4];
  if ((threadIdx.x & 63) == 0) part[threadIdx.x >> 6] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) t += part[w];
    out[n] = t;
  }
}

torch::Tensor colsum(torch::Tensor g) {
  CHECK_GPU(g);
  auto gc = g.to(torch::kFloat32).contiguous();
  int64_t M = gc.size(0), N = gc.size(1);
  auto out = torch::empty({N}, gc.options());
  hipLaunchKernelGGL(colsum_kernel, dim3((unsigned)N), dim3(ELEM_BLOCK),
                     0, current_stream(), gc.data_ptr<float>(),
                     out.data_ptr<float>(), M, N);
  return out;
}

// K8 — fused Gram-Schmidt column orthogonalization (PowerSGD).
// One workgroup per matrix [n, r] (r small, typically 1-4): each column
// pass computes the norm (wave-reduce over rows), scales, and projects the
// remaining columns — a single launch instead of 3*r elementwise kernels.
__global__ void gram_schmidt_kernel(float* __restrict__ m, int n, int r,
                                    float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem_gs[];
  float* red = reinterpret_cast<float*>(smem_gs);  // [blockDim/64]
  const int tid = threadIdx.x;
  const int nw = blockDim.x / WAVE_SIZE;
  for (int i = 0; i < r; ++i) {
    // norm of column i
    float s = 0.f;
    for (int row = tid; row < n; row += blockDim.x) {
      float v = m[(int64_t)row * r + i];
      s += v * v;
    }
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
      s += __shfl_down(s, off);
    if ((tid & 63) == 0) red[tid >> 6] = s;
    __syncthreads();
    if (tid == 0) {
      float t = 0.f;
      for (int w = 0; w < nw; ++w) t += red[w];
      red[0] = sqrtf(t) + eps;
    }
    __syncthreads();
    const float inv = 1.0f / red[0];
    for (int row = tid; row < n; row += blockDim.x)
      m[(int64_t)row * r + i] *= inv;
    __syncthreads();
    // project out of the remaining columns
    for (int j = i + 1; j < r; ++j) {
      float d = 0.f;
      for (int row = tid; row < n; row += blockDim.x)
        d += m[(int64_t)row * r + i] * m[(int64_t)row * r + j];
      for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
        d += __shfl_down(d, off);
      if ((tid & 63) == 0) red[tid >> 6] = d;
      __syncthreads();
      if (tid == 0) {
        float t = 0.f;
        for (int w = 0; w < nw; ++w) t += red[w];
        red[0] = t;
      }
      __syncthreads();
      const float dot = red[0];
      for (int row = tid; row < n; row += blockDim.x)
        m[(int64_t)row * r + j] -= dot * m[(int64_t)row * r + i];
      __syncthreads();
    }
  }
}

void gram_schmidt(torch::Tensor m, double eps) {
  CHECK_GPU(m); CHECK_CONTIG(m);
  TORCH_CHECK(m.dim() == 2 && m.scalar_type() == torch::kFloat32);
  int n = (int)m.size(0), r = (int)m.size(1);
  hipLaunchKernelGGL(gram_schmidt_kernel, dim3(1), dim3(256),
                     16 * sizeof(float), current_stream(),
                     m.data_ptr<float>(), n, r, (float)eps);
}
