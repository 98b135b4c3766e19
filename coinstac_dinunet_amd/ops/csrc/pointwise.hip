// Pointwise (1x1) convolution kernels — 2D and 3D channel-mixing convs
// (ResNet downsamples, the UNet3D segmentation head). A 1x1 conv in
// channels-first layout is, per batch element, a plain GEMM over the
// channel dim:
//   fwd  : out_n[Co, S] = W[Co, Ci] @ x_n[Ci, S]
//   dgrad: gx_n[Ci, S]  = W^T       @ go_n[Co, S]
//   wgrad: gw[Co, Ci]   = sum_n go_n @ x_n^T
// All three run as batched MFMA GEMMs (weights shared across the batch,
// grid = row-tiles x col-tiles x batch). An earlier streaming form that
// parallelized only over spatial positions collapsed to a handful of
// blocks at ResNet downsample shapes (S as small as 49) and left the
// chip ~1% occupied (r2 profile: 629 of 774 ms) — see git history.
#include "common.h"

// out[n,co,s] (+= bias) from x[n,ci,s]; TRANSPOSE_W=true computes
// dgrad: gx[n,ci,s] = sum_co w[co,ci] * go[n,co,s] with Cin/Cout roles
// swapped by the caller (then "cin" here is Cout of the conv).
// Each thread accumulates 4 spatial points (spaced blockDim apart so each
// of the 4 loads per ci stays coalesced) — 4 FMAs per LDS weight read
// instead of 1; co_chunk <= 16 keeps acc in 64 VGPRs.
// ---------------------------------------------------------------------------
// Batched MFMA GEMM: C_n[i, j] = sum_c opA(i, c) * B_n[c, j], weights
// shared across the batch, grid.z = n.
// ---------------------------------------------------------------------------
#include <hip/hip_bf16.h>
typedef __attribute__((ext_vector_type(8))) __bf16 pwbf16x8;
typedef __attribute__((ext_vector_type(4))) float pwf32x4;
#define PWBM 64
#define PWBN 64
#define PWBK 16
#define PWLDK (PWBK + 1)

template <bool TRANS_A, bool HAS_BIAS>
__global__ __launch_bounds__(256) void pw_gemm_batched_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    const float* __restrict__ bias, __bf16* __restrict__ C, int M, int N,
    int K, int64_t strideB, int64_t strideC) {
  __shared__ float sA[PWBM][PWLDK];
  __shared__ float sB[PWBN][PWLDK];

  const int bm = blockIdx.x * PWBM;
  const int bn = blockIdx.y * PWBN;
  const __bf16* Bn = B + (int64_t)blockIdx.z * strideB;
  __bf16* Cn = C + (int64_t)blockIdx.z * strideC;
  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int wm = (wave >> 1) * 32;
  const int wn = (wave & 1) * 32;

  pwf32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += PWBK) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int idx = tid + r * 256;  // 0..1023 = PWBM*PWBK
      if (TRANS_A) {
        int c = idx / PWBM, row = idx % PWBM;
        int gm = bm + row, gk = k0 + c;
        sA[row][c] = (gm < M && gk < K)
                         ? __bfloat162float(A[(int64_t)gk * M + gm])
                         : 0.f;
      } else {
        int row = idx / PWBK, col = idx % PWBK;
        int gm = bm + row, gk = k0 + col;
        sA[row][col] = (gm < M && gk < K)
                           ? __bfloat162float(A[(int64_t)gm * K + gk])
                           : 0.f;
      }
      // B[c, j]: coalesced along j, stored transposed
      int c = idx / PWBN, col = idx % PWBN;
      int gn = bn + col, gk = k0 + c;
      sB[col][c] = (gn < N && gk < K)
                       ? __bfloat162float(Bn[(int64_t)gk * N + gn])
                       : 0.f;
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < PWBK; kk += 4) {
      const int ar = lane & 15, ak = lane >> 4;
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          float a = sA[wm + i * 16 + ar][kk + ak];
          float b = sB[wn + j * 16 + ar][kk + ak];
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[i][j],
                                                           0, 0, 0);
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
          float v = acc[i][j][r];
          if (HAS_BIAS) v += bias[gm];  // bias indexed by output channel
          Cn[(int64_t)gm * N + gn] = (__bf16)v;
        }
      }
}

// TRANS_A note: fwd uses A = W [Co, Ci] row-major (opA(i,c) = A[i*K+c],
// TRANS_A=false); dgrad uses the SAME W but needs opA(i=ci, c=co) =
// W[co*Ci+ci] = A[c*M + i] — the TRANS_A=true indexing above.

// wgrad as batched MFMA: gw[Co, Ci] = sum_n go_n[Co, S] @ x_n[Ci, S]^T.
// Per z-block NZ batches are accumulated in registers, then one fp32
// atomicAdd per tile element folds partials across z-blocks (replaces
// the streaming pw_wgrad whose per-(n,s) atomics serialized at ~1 ms per
// call on the 512x256 downsample shapes).
template <int NZ>
__global__ __launch_bounds__(256) void pw_wgrad_batched_kernel(
    const __bf16* __restrict__ go, const __bf16* __restrict__ x,
    float* __restrict__ gw, int M, int N, int K, int batches,
    int64_t strideA, int64_t strideB) {
  __shared__ float sA[PWBM][PWLDK];
  __shared__ float sB[PWBN][PWLDK];

  const int bm = blockIdx.x * PWBM;
  const int bn = blockIdx.y * PWBN;
  const int n0 = blockIdx.z * NZ;
  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int wm = (wave >> 1) * 32;
  const int wn = (wave & 1) * 32;

  pwf32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int nz = 0; nz < NZ && n0 + nz < batches; ++nz) {
    const __bf16* An = go + (int64_t)(n0 + nz) * strideA;
    const __bf16* Bn = x + (int64_t)(n0 + nz) * strideB;
    for (int k0 = 0; k0 < K; k0 += PWBK) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int idx = tid + r * 256;
        int row = idx / PWBK, col = idx % PWBK;
        int gm = bm + row, gk = k0 + col;
        sA[row][col] = (gm < M && gk < K)
                           ? __bfloat162float(An[(int64_t)gm * K + gk])
                           : 0.f;
        int gn = bn + row;
        sB[row][col] = (gn < N && gk < K)
                           ? __bfloat162float(Bn[(int64_t)gn * K + gk])
                           : 0.f;
      }
      __syncthreads();
#pragma unroll
      for (int kk = 0; kk < PWBK; kk += 4) {
        const int ar = lane & 15, ak = lane >> 4;
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
          for (int j = 0; j < 2; ++j) {
            float a = sA[wm + i * 16 + ar][kk + ak];
            float b = sB[wn + j * 16 + ar][kk + ak];
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                a, b, acc[i][j], 0, 0, 0);
          }
      }
      __syncthreads();
    }
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
        if (gm < M && gn < N)
          atomicAdd(&gw[(int64_t)gm * N + gn], acc[i][j][r]);
      }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

torch::Tensor conv3d_pw_fwd(torch::Tensor x, torch::Tensor w,
                            torch::Tensor bias) {
  CHECK_GPU(x);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "pw_fwd wants bf16 x");
  auto xc = x.contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  const int N = (int)xc.size(0), Cin = (int)xc.size(1);
  const int Cout = (int)wc.size(0);
  const int64_t S = xc.numel() / ((int64_t)N * Cin);
  TORCH_CHECK(wc.numel() == (int64_t)Cout * Cin, "1x1x1 weight expected");
  auto sizes = xc.sizes().vec();
  sizes[1] = Cout;
  auto out = torch::empty(sizes, xc.options());
  const float* bp = nullptr;
  torch::Tensor bc;
  if (bias.defined() && bias.numel() > 0) {
    bc = bias.to(torch::kFloat32).contiguous();
    bp = bc.data_ptr<float>();
  }
  dim3 grid((Cout + PWBM - 1) / PWBM, (unsigned)((S + PWBN - 1) / PWBN),
            N);
  if (bp) {
    hipLaunchKernelGGL((pw_gemm_batched_kernel<false, true>), grid,
                       dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(wc.data_ptr()),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()), bp,
                       reinterpret_cast<__bf16*>(out.data_ptr()), Cout,
                       (int)S, Cin, (int64_t)Cin * S, (int64_t)Cout * S);
  } else {
    hipLaunchKernelGGL((pw_gemm_batched_kernel<false, false>), grid,
                       dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(wc.data_ptr()),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       nullptr,
                       reinterpret_cast<__bf16*>(out.data_ptr()), Cout,
                       (int)S, Cin, (int64_t)Cin * S, (int64_t)Cout * S);
  }
  return out;
}

torch::Tensor conv3d_pw_dgrad(torch::Tensor go, torch::Tensor w) {
  CHECK_GPU(go);
  auto g = go.to(torch::kBFloat16).contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  const int N = (int)g.size(0), Cout = (int)g.size(1);
  const int Cin = (int)wc.size(1);
  const int64_t S = g.numel() / ((int64_t)N * Cout);
  auto sizes = g.sizes().vec();
  sizes[1] = Cin;
  auto gx = torch::empty(sizes, g.options());
  dim3 grid((Cin + PWBM - 1) / PWBM, (unsigned)((S + PWBN - 1) / PWBN), N);
  hipLaunchKernelGGL((pw_gemm_batched_kernel<true, false>), grid, dim3(256),
                     0, current_stream(),
                     reinterpret_cast<const __bf16*>(wc.data_ptr()),
                     reinterpret_cast<const __bf16*>(g.data_ptr()), nullptr,
                     reinterpret_cast<__bf16*>(gx.data_ptr()), Cin, (int)S,
                     Cout, (int64_t)Cout * S, (int64_t)Cin * S);
  return gx;
}

torch::Tensor conv3d_pw_wgrad(torch::Tensor x, torch::Tensor go) {
  CHECK_GPU(x);
  auto xc = x.to(torch::kBFloat16).contiguous();
  auto g = go.to(torch::kBFloat16).contiguous();
  const int N = (int)xc.size(0), Cin = (int)xc.size(1);
  const int Cout = (int)g.size(1);
  const int64_t S = xc.numel() / ((int64_t)N * Cin);
  auto gw = torch::zeros({Cout, Cin},
                         xc.options().dtype(torch::kFloat32));
  // batched MFMA: per n, go_n[Co,S] @ x_n[Ci,S]^T, NZ batches per block
  constexpr int NZ = 8;
  dim3 grid((Cout + PWBM - 1) / PWBM, (Cin + PWBN - 1) / PWBN,
            (N + NZ - 1) / NZ);
  hipLaunchKernelGGL((pw_wgrad_batched_kernel<NZ>), grid, dim3(256), 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(g.data_ptr()),
                     reinterpret_cast<const __bf16*>(xc.data_ptr()),
                     gw.data_ptr<float>(), Cout, Cin, (int)S, N,
                     (int64_t)Cout * S, (int64_t)Cin * S);
  return gw;
}
