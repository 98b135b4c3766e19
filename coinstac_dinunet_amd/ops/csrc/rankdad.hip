// K10 — fused deflated power iteration for rankDAD compression.
// Reference shape: /root/reference/coinstac_dinunet/distrib/rankdad/spi.py:9-86
// (power_iteration_BC), which our python port in distrib/rankdad.py mirrors.
// The torch/rocBLAS chain runs ~rank*iters tiny matvec launches per layer
// per round (~50 for rank=10, iters=5); at rankDAD's layer shapes
// (n,m ~ 1e2-1e3, k = batch) every launch is microseconds of work, so the
// round is launch-latency bound. This kernel runs the WHOLE extraction —
// matvec chain, deflation, norms, sigma, c-vectors — in ONE launch on one
// 1024-thread workgroup (16 waves), with the Gram precompute (a real GEMM)
// done by the in-tree MFMA linear kernels on the host side.
//
// Two algebraic branches, as in the reference:
//   small-k (k <= m): iterate v = G2 @ b with G2 = (B C^T)(B C^T)^T [n,n];
//     sigma = ||BCT^T b||, c = BCT^T b / sigma.
//   big-k: iterate v = B (CC (B^T b)) with CC = C^T C [k,k];
//     sigma = sqrt(Bv . CC Bv), c = C Bv / sigma.
// Unit b-vectors are stored in Bf rows during extraction (deflation reads
// them) and scaled by sigma at the end, matching the reference's
// Bf = stack(sigma_i * b_i).
#include "common.h"

#define PIB_THREADS 1024
#define PIB_MAX_RANK 32

// ---- block-wide helpers (uniform control flow; all contain barriers) ----
__device__ static float pib_reduce(float v, float* red) {
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off);
  const int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) red[wave] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) t += red[w];
    red[16] = t;
  }
  __syncthreads();
  const float t = red[16];
  __syncthreads();
  return t;
}

__device__ static float pib_dot(const float* x, const float* y, int len,
                                float* red) {
  float s = 0.f;
  for (int i = threadIdx.x; i < len; i += blockDim.x) s += x[i] * y[i];
  return pib_reduce(s, red);
}

// y[r] = sum_j M[r,j] x[j]  (wave per row, coalesced row reads)
__device__ static void pib_matvec(const float* M, const float* x, float* y,
                                  int rows, int cols) {
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int nw = blockDim.x >> 6;
  for (int r = wave; r < rows; r += nw) {
    float s = 0.f;
    for (int j = lane; j < cols; j += WAVE_SIZE)
      s += M[(int64_t)r * cols + j] * x[j];
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
      s += __shfl_down(s, off);
    if (lane == 0) y[r] = s;
  }
  __syncthreads();
}

// y[c] = sum_i M[i,c] x[i]  (wave per column; strided loads — fine at the
// small m these layers have)
__device__ static void pib_matvec_t(const float* M, const float* x, float* y,
                                    int rows, int cols) {
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int nw = blockDim.x >> 6;
  for (int c = wave; c < cols; c += nw) {
    float s = 0.f;
    for (int i = lane; i < rows; i += WAVE_SIZE)
      s += M[(int64_t)i * cols + c] * x[i];
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
      s += __shfl_down(s, off);
    if (lane == 0) y[c] = s;
  }
  __syncthreads();
}

__global__ __launch_bounds__(PIB_THREADS) void power_iter_bc_kernel(
    const float* __restrict__ G2,   // [n,n]  (small-k) or null
    const float* __restrict__ BCT,  // [n,m]  (small-k) or null
    const float* __restrict__ B,    // [n,k]  (big-k) or null
    const float* __restrict__ CC,   // [k,k]  (big-k) or null
    const float* __restrict__ Cm,   // [m,k]  (big-k) or null
    const float* __restrict__ starts,  // [rank, n]
    float* __restrict__ Bf,  // [rank, n] rows = components
    float* __restrict__ Cf,  // [rank, m]
    float* __restrict__ scr,  // b[n] | v[n] | t1[max(k,m)] | t2[k]
    int* __restrict__ ncomp_out, int n, int m, int k, int rank, int iters,
    float tol) {
  __shared__ float red[17];
  const bool small_k = (G2 != nullptr);
  float* b = scr;
  float* v = scr + n;
  float* t1 = scr + 2 * n;
  float* t2 = t1 + (k > m ? k : m);

  float sigs[PIB_MAX_RANK];
  int ncomp = 0;
  float lam1 = -1.f;

  for (int rr = 0; rr < rank; ++rr) {
    for (int i = threadIdx.x; i < n; i += blockDim.x)
      b[i] = starts[(int64_t)rr * n + i];
    __syncthreads();

    bool degenerate = false;
    float norm = 0.f;
    for (int it = 0; it < iters; ++it) {
      if (small_k) {
        pib_matvec(G2, b, v, n, n);
      } else {
        pib_matvec_t(B, b, t1, n, k);   // t1 = B^T b      [k]
        pib_matvec(CC, t1, t2, k, k);   // t2 = CC t1      [k]
        pib_matvec(B, t2, v, n, k);     // v  = B t2       [n]
      }
      for (int j = 0; j < ncomp; ++j) {  // deflate
        const float* bb = Bf + (int64_t)j * n;
        const float d = pib_dot(bb, v, n, red);
        for (int i = threadIdx.x; i < n; i += blockDim.x) v[i] -= bb[i] * d;
        __syncthreads();
      }
      norm = sqrtf(pib_dot(v, v, n, red));
      if ((lam1 >= 0.f && norm < tol * lam1) || norm < 1e-12f) {
        degenerate = true;
        break;
      }
      const float inv = 1.f / norm;
      for (int i = threadIdx.x; i < n; i += blockDim.x) b[i] = v[i] * inv;
      __syncthreads();
    }
    if (degenerate) break;

    for (int j = 0; j < ncomp; ++j) {  // re-orthogonalize before sigma
      const float* bb = Bf + (int64_t)j * n;
      const float d = pib_dot(bb, b, n, red);
      for (int i = threadIdx.x; i < n; i += blockDim.x) b[i] -= bb[i] * d;
      __syncthreads();
    }
    const float bn = sqrtf(pib_dot(b, b, n, red));
    if (bn < 1e-12f) break;
    const float binv = 1.f / bn;
    for (int i = threadIdx.x; i < n; i += blockDim.x) b[i] *= binv;
    __syncthreads();
    if (lam1 < 0.f) lam1 = norm;

    float sigma;
    if (small_k) {
      pib_matvec_t(BCT, b, t1, n, m);  // t1 = BCT^T b = C B^T b  [m]
      sigma = sqrtf(pib_dot(t1, t1, m, red));
    } else {
      pib_matvec_t(B, b, t1, n, k);    // Bv              [k]
      pib_matvec(CC, t1, t2, k, k);    // CC Bv           [k]
      sigma = sqrtf(fmaxf(pib_dot(t1, t2, k, red), 0.f));
    }
    if (isnan(sigma) || sigma < 1e-12f || sigma * sigma < tol * lam1) break;

    const float sinv = 1.f / sigma;
    if (small_k) {
      for (int i = threadIdx.x; i < m; i += blockDim.x)
        Cf[(int64_t)rr * m + i] = t1[i] * sinv;
    } else {
      pib_matvec(Cm, t1, t2, m, k);    // C Bv            [m]
      for (int i = threadIdx.x; i < m; i += blockDim.x)
        Cf[(int64_t)rr * m + i] = t2[i] * sinv;
    }
    for (int i = threadIdx.x; i < n; i += blockDim.x)
      Bf[(int64_t)rr * n + i] = b[i];
    __syncthreads();
    sigs[ncomp] = sigma;
    ++ncomp;
  }

  // scale unit b rows by their singular values
  for (int j = 0; j < ncomp; ++j) {
    const float s = sigs[j];
    for (int i = threadIdx.x; i < n; i += blockDim.x)
      Bf[(int64_t)j * n + i] *= s;
  }
  if (threadIdx.x == 0) *ncomp_out = ncomp;
}

// host entry — Gram precompute on the in-tree MFMA linear kernels
torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor weight,
                         torch::Tensor bias, bool relu);
torch::Tensor linear_wgrad(torch::Tensor go, torch::Tensor x);

std::vector<torch::Tensor> power_iter_bc(torch::Tensor B, torch::Tensor C,
                                         int64_t rank, int64_t iters,
                                         double tol, torch::Tensor starts) {
  CHECK_GPU(B);
  TORCH_CHECK(B.scalar_type() == torch::kFloat32 &&
              C.scalar_type() == torch::kFloat32, "power_iter_bc: fp32 only");
  TORCH_CHECK(rank >= 1 && rank <= PIB_MAX_RANK, "rank must be 1..32");
  auto Bc = B.contiguous();
  auto Cc = C.contiguous();
  const int n = (int)Bc.size(0), k = (int)Bc.size(1);
  const int m = (int)Cc.size(0);
  TORCH_CHECK(Cc.size(1) == k, "B/C k mismatch");
  TORCH_CHECK(starts.size(0) == rank && starts.size(1) == n,
              "starts must be [rank, n]");
  auto st = starts.to(Bc.options()).contiguous();
  auto empty = torch::empty({0}, Bc.options());

  const bool small_k = k <= m;
  torch::Tensor G2, BCT, CCt;
  if (small_k) {
    BCT = linear_fwd(Bc, Cc, empty, false);   // B C^T    [n,m]
    G2 = linear_fwd(BCT, BCT, empty, false);  // (BC^T)(BC^T)^T [n,n]
  } else {
    CCt = linear_wgrad(Cc, Cc);               // C^T C    [k,k]
  }

  auto Bf = torch::zeros({rank, n}, Bc.options());
  auto Cf = torch::zeros({rank, m}, Bc.options());
  auto scr = torch::empty({2 * (int64_t)n + std::max(k, m) + k},
                          Bc.options());
  auto ncomp = torch::zeros({1}, Bc.options().dtype(torch::kInt32));

  hipLaunchKernelGGL(
      power_iter_bc_kernel, dim3(1), dim3(PIB_THREADS), 0, current_stream(),
      small_k ? G2.data_ptr<float>() : nullptr,
      small_k ? BCT.data_ptr<float>() : nullptr,
      small_k ? nullptr : Bc.data_ptr<float>(),
      small_k ? nullptr : CCt.data_ptr<float>(),
      small_k ? nullptr : Cc.data_ptr<float>(), st.data_ptr<float>(),
      Bf.data_ptr<float>(), Cf.data_ptr<float>(), scr.data_ptr<float>(),
      ncomp.data_ptr<int>(), n, m, k, (int)rank, (int)iters, (float)tol);
  return {Bf, Cf, ncomp};
}

// K11 helper — per-row sums of a [out, r] factor:
// out[i] = sum_r M[i, r] (the reconstructed layer's bias gradient).
__global__ void rowsum_kernel(const float* __restrict__ M,
                              float* __restrict__ out, int64_t rows,
                              int64_t cols) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < rows;
       i += (int64_t)gridDim.x * blockDim.x) {
    float s = 0.f;
    for (int64_t j = 0; j < cols; ++j) s += M[i * cols + j];
    out[i] = s;
  }
}

torch::Tensor rowsum(torch::Tensor m) {
  CHECK_GPU(m);
  auto mc = m.to(torch::kFloat32).contiguous();
  const int64_t rows = mc.size(0), cols = mc.size(1);
  auto out = torch::empty({rows}, mc.options());
  hipLaunchKernelGGL(rowsum_kernel, dim3(elem_grid(rows, 1)),
                     dim3(ELEM_BLOCK), 0, current_stream(),
                     mc.data_ptr<float>(), out.data_ptr<float>(), rows, cols);
  return out;
}
