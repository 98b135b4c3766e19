// K1-2D — implicit-GEMM Conv2d (3x3, pad 1, stride 1/2) on the matrix
// cores, for the ResNet-18/FedAvg config (BASELINE.json #5). Round 1 ran
// every ResNet FLOP through MIOpen; this gives the 2D
// conv family the same in-tree implicit-GEMM treatment as conv3d.hip,
// adapted by dropping the depth axis (9 taps, (n,oh,ow) position decode).
//
// GEMM views (NCHW, w-fastest so gathers coalesce):
//   FWD  : C[M=N*OH*OW, Cout] = patch(x)[M, Cin*9] @ W[Cout, Cin*9]^T
//   DGRAD: C[M=N*H*W,  Cin ] = gather(go)[M, Cout*9] @ W' (flipped taps;
//          stride-2 via divisibility mask — the 2D parity decomposition is
//          a later optimization, the masked form is correct at 1/4 density)
//   WGRAD: C[Cout, Cin*9] = go^T @ patch(x)  (split-K over m, fp32 atomics)
// Fragment layout identical to conv3d.hip (mfma_f32_16x16x32_bf16).
#include "common.h"

#include <hip/hip_bf16.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

struct Conv2dDims {
  int N, Cin, H, W;
  int Cout, OH, OW;
  int stride;  // pad fixed at 1, kernel 3x3
};

#define CBM2 64
#define CBN2 64
#define CBK2 32
#define LDA_PAD2 8

template <bool DGRAD>
__device__ inline __bf16 gather2_w(const __bf16* __restrict__ w,
                                   const Conv2dDims& cd, int k, int col) {
  if (!DGRAD) {
    if (col >= cd.Cout || k >= cd.Cin * 9) return (__bf16)0.f;
    return w[(int64_t)col * (cd.Cin * 9) + k];
  }
  const int co = k / 9;
  const int r = k - co * 9;
  if (co >= cd.Cout || col >= cd.Cin) return (__bf16)0.f;
  // taps un-flipped here; the (ih + 1 - kh) mapping in the A gather
  // implements the transposed conv (same convention as conv3d.hip)
  return w[((int64_t)co * cd.Cin + col) * 9 + r];
}

template <bool DGRAD, int STRIDE, bool FUSE_BN = false>
__global__ __launch_bounds__(256) void conv2d_igemm_kernel(
    const __bf16* __restrict__ Ain, const __bf16* __restrict__ w,
    __bf16* __restrict__ out, Conv2dDims cd, int64_t M, int Ncol, int K,
    const float* __restrict__ bn_ab) {
  __shared__ __bf16 sA[CBM2][CBK2 + LDA_PAD2];
  __shared__ __bf16 sBT[CBN2][CBK2 + LDA_PAD2];

  const int64_t bm = (int64_t)blockIdx.x * CBM2;
  const int bn = blockIdx.y * CBN2;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wm = (wave >> 1) * 32, wn = (wave & 1) * 32;

  const int kk_t = tid >> 3;
  const int mbase = (tid * 8) & 63;
  const int SH = DGRAD ? cd.H : cd.OH;
  const int SW = DGRAD ? cd.W : cd.OW;
  int pn[8], ph[8], pw[8];
  {
    int64_t m = bm + mbase;
    int ww = (int)(m % SW);
    int64_t t = m / SW;
    int hh = (int)(t % SH);
    int nn = (int)(t / SH);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      pn[j] = nn; ph[j] = hh; pw[j] = ww;
      if (++ww == SW) { ww = 0; if (++hh == SH) { hh = 0; ++nn; } }
    }
  }
  const bool m_ok = (bm + mbase + 7) < M;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)cd.H * cd.W;
  const int64_t OHW = (int64_t)cd.OH * cd.OW;

  for (int k0 = 0; k0 < K; k0 += CBK2) {
    const int k = k0 + kk_t;
    if (!DGRAD) {
      const int ci = k / 9;
      const int r = k - ci * 9;
      const int kh = r / 3, kw = r % 3;
      const bool k_ok = ci < cd.Cin;
      float a_c = 1.f, b_c = 0.f;
      if (FUSE_BN && k_ok) { a_c = bn_ab[ci * 2]; b_c = bn_ab[ci * 2 + 1]; }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __bf16 v = (__bf16)0.f;
        if (k_ok && (m_ok || (bm + mbase + j) < M)) {
          const int ih = ph[j] * STRIDE - 1 + kh;
          const int iw = pw[j] * STRIDE - 1 + kw;
          if ((unsigned)ih < (unsigned)cd.H && (unsigned)iw < (unsigned)cd.W) {
            v = Ain[((int64_t)pn[j] * cd.Cin + ci) * HW +
                    (int64_t)ih * cd.W + iw];
            if (FUSE_BN) v = (__bf16)fmaxf(a_c * (float)v + b_c, 0.f);
          }
        }
        sA[mbase + j][kk_t] = v;
      }
    } else {
      const int co = k / 9;
      const int r = k - co * 9;
      const int kh = r / 3, kw = r % 3;
      const bool k_ok = co < cd.Cout;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __bf16 v = (__bf16)0.f;
        if (k_ok && (m_ok || (bm + mbase + j) < M)) {
          const int th = ph[j] + 1 - kh, tw = pw[j] + 1 - kw;
          if (STRIDE == 1 ||
              (!(th & (STRIDE - 1)) && !(tw & (STRIDE - 1)))) {
            const int oh = th / STRIDE, ow = tw / STRIDE;
            if ((unsigned)oh < (unsigned)cd.OH &&
                (unsigned)ow < (unsigned)cd.OW)
              v = Ain[((int64_t)pn[j] * cd.Cout + co) * OHW +
                      (int64_t)oh * cd.OW + ow];
          }
        }
        sA[mbase + j][kk_t] = v;
      }
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      int idx = tid * 8 + e;
      int kk = idx & 31, col = idx >> 5;
      sBT[col][kk] = gather2_w<DGRAD>(w, cd, k0 + kk, bn + col);
    }
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < CBK2; ks += 32) {
      const int row = lane & 15, kg = lane >> 4;
      bf16x8 afrag[2], bfrag[2];
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          afrag[i][j] = sA[wm + i * 16 + row][ks + kg * 8 + j];
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          bfrag[i][j] = sBT[wn + i * 16 + row][ks + kg * 8 + j];
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  const int64_t spatial = DGRAD ? (int64_t)cd.H * cd.W
                                : (int64_t)cd.OH * cd.OW;
  const int nch = DGRAD ? cd.Cin : cd.Cout;
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int64_t m = bm + wm + i * 16 + crow0 + r;
        int col = bn + wn + j * 16 + ccol;
        if (m < M && col < nch) {
          int64_t n = m / spatial, sp = m % spatial;
          out[((int64_t)n * nch + col) * spatial + sp] =
              (__bf16)(acc[i][j][r]);
        }
      }
}

// ---------------------------------------------------------------------------
// WGRAD: split-K implicit GEMM with fp32 atomics (structure of
// conv3d.hip's conv3d_wgrad_kernel with the depth axis dropped).
// ---------------------------------------------------------------------------
#define WMB2 128

template <bool FUSE_BN = false>
__global__ __launch_bounds__(256) void conv2d_wgrad_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ go,
    float* __restrict__ dw, Conv2dDims cd, int64_t M, int K, int64_t chunk,
    const float* __restrict__ bn_ab) {
  __shared__ __bf16 sGoT[32][WMB2 + LDA_PAD2];
  __shared__ __bf16 sXT[32][WMB2 + LDA_PAD2];

  const int co0 = blockIdx.x * 32;
  const int k0 = blockIdx.y * 32;
  const int64_t m0 = (int64_t)blockIdx.z * chunk;
  const int64_t mEnd = min(m0 + chunk, M);
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wi = wave >> 1, wj = wave & 1;

  const int mi0 = (tid & 7) * 4;
  const int ct = tid >> 3;
  const int kq = k0 + ct;
  const int ci = kq / 9;
  const int rr = kq - ci * 9;
  const int kh = rr / 3, kw = rr % 3;
  const bool k_ok = ci < cd.Cin && kq < K;
  const bool c_ok = (co0 + ct) < cd.Cout;
  float a_c = 1.f, b_c = 0.f;
  if (FUSE_BN && k_ok) { a_c = bn_ab[ci * 2]; b_c = bn_ab[ci * 2 + 1]; }

  int nn, oh, ow;
  {
    int64_t m = m0 + mi0;
    ow = (int)(m % cd.OW);
    int64_t t = m / cd.OW;
    oh = (int)(t % cd.OH);
    nn = (int)(t / cd.OH);
  }

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int64_t spatial = (int64_t)cd.OH * cd.OW;
  const int64_t HW = (int64_t)cd.H * cd.W;

  for (int64_t mb = m0; mb < mEnd; mb += WMB2) {
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      int jn = nn, jh = oh, jw = ow;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int mloc = q * 32 + mi0 + j;
        const int64_t m = mb + mloc;
        const bool ok = m < mEnd;
        __bf16 gv = (__bf16)0.f;
        if (ok && c_ok)
          gv = go[((int64_t)jn * cd.Cout + (co0 + ct)) * spatial +
                  (int64_t)jh * cd.OW + jw];
        sGoT[ct][mloc] = gv;
        __bf16 xv = (__bf16)0.f;
        if (ok && k_ok) {
          const int ih = jh * cd.stride - 1 + kh;
          const int iw = jw * cd.stride - 1 + kw;
          if ((unsigned)ih < (unsigned)cd.H &&
              (unsigned)iw < (unsigned)cd.W) {
            xv = x[((int64_t)jn * cd.Cin + ci) * HW +
                   (int64_t)ih * cd.W + iw];
            if (FUSE_BN) xv = (__bf16)fmaxf(a_c * (float)xv + b_c, 0.f);
          }
        }
        sXT[ct][mloc] = xv;
        if (++jw == cd.OW) { jw = 0; if (++jh == cd.OH) { jh = 0; ++jn; } }
      }
      ow += 32;
      while (ow >= cd.OW) {
        ow -= cd.OW;
        if (++oh == cd.OH) { oh = 0; ++nn; }
      }
    }
    __syncthreads();

    const int row = lane & 15, kg = lane >> 4;
#pragma unroll
    for (int ks = 0; ks < WMB2; ks += 32) {
      bf16x8 afrag, bfrag;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        afrag[j] = sGoT[wi * 16 + row][ks + kg * 8 + j];
        bfrag[j] = sXT[wj * 16 + row][ks + kg * 8 + j];
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc,
                                                    0, 0, 0);
    }
    __syncthreads();
  }

  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int co = co0 + wi * 16 + crow0 + r;
    int k = k0 + wj * 16 + ccol;
    if (co < cd.Cout && k < K)
      atomicAdd(&dw[(int64_t)co * K + k], acc[r]);
  }
}

// ---------------------------------------------------------------------------
// hosts
// ---------------------------------------------------------------------------
static Conv2dDims make_dims2(const torch::Tensor& x, const torch::Tensor& w,
                             int stride) {
  Conv2dDims cd;
  cd.N = (int)x.size(0); cd.Cin = (int)x.size(1);
  cd.H = (int)x.size(2); cd.W = (int)x.size(3);
  cd.Cout = (int)w.size(0);
  cd.stride = stride;
  cd.OH = (cd.H + 2 - 3) / stride + 1;
  cd.OW = (cd.W + 2 - 3) / stride + 1;
  return cd;
}

torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, int64_t stride,
                         c10::optional<torch::Tensor> bn_ab_opt) {
  torch::Tensor bn_ab = bn_ab_opt.value_or(torch::Tensor());
  CHECK_GPU(x);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "conv2d_fwd wants bf16");
  auto xc = x.contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  auto cd = make_dims2(xc, wc, (int)stride);
  TORCH_CHECK(wc.size(2) == 3 && wc.size(3) == 3 && wc.size(1) == cd.Cin,
              "3x3 kernels only");
  TORCH_CHECK(stride == 1 || stride == 2, "stride must be 1 or 2");
  const bool fuse = bn_ab.defined() && bn_ab.numel() > 0;
  torch::Tensor ab;
  const float* abp = nullptr;
  if (fuse) {
    ab = bn_ab.to(torch::kFloat32).contiguous();
    TORCH_CHECK(ab.numel() == 2 * cd.Cin, "bn_ab must be [Cin,2]");
    abp = ab.data_ptr<float>();
  }
  auto out = torch::empty({cd.N, cd.Cout, cd.OH, cd.OW}, xc.options());
  int64_t M = (int64_t)cd.N * cd.OH * cd.OW;
  int K = cd.Cin * 9;
  dim3 grid((unsigned)((M + CBM2 - 1) / CBM2), (cd.Cout + CBN2 - 1) / CBN2);
  auto L = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<const __bf16*>(wc.data_ptr()),
                       reinterpret_cast<__bf16*>(out.data_ptr()), cd, M,
                       cd.Cout, K, abp);
  };
  if (fuse) {
    if (stride == 1) L(conv2d_igemm_kernel<false, 1, true>);
    else L(conv2d_igemm_kernel<false, 2, true>);
  } else if (stride == 1) {
    L(conv2d_igemm_kernel<false, 1>);
  } else {
    L(conv2d_igemm_kernel<false, 2>);
  }
  return out;
}

torch::Tensor conv2d_dgrad(torch::Tensor go, torch::Tensor w,
                           std::vector<int64_t> in_shape, int64_t stride) {
  CHECK_GPU(go);
  auto g = go.to(torch::kBFloat16).contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  Conv2dDims cd;
  cd.N = (int)in_shape[0]; cd.Cin = (int)in_shape[1];
  cd.H = (int)in_shape[2]; cd.W = (int)in_shape[3];
  cd.Cout = (int)wc.size(0);
  cd.stride = (int)stride;
  cd.OH = (int)g.size(2); cd.OW = (int)g.size(3);
  TORCH_CHECK(stride == 1 || stride == 2, "stride must be 1 or 2");
  auto dx = torch::empty(in_shape, g.options());
  int64_t M = (int64_t)cd.N * cd.H * cd.W;
  int K = cd.Cout * 9;
  dim3 grid((unsigned)((M + CBM2 - 1) / CBM2), (cd.Cin + CBN2 - 1) / CBN2);
  auto L = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(g.data_ptr()),
                       reinterpret_cast<const __bf16*>(wc.data_ptr()),
                       reinterpret_cast<__bf16*>(dx.data_ptr()), cd, M,
                       cd.Cin, K, (const float*)nullptr);
  };
  if (stride == 1) L(conv2d_igemm_kernel<true, 1>);
  else L(conv2d_igemm_kernel<true, 2>);
  return dx;
}

torch::Tensor conv2d_wgrad(torch::Tensor x, torch::Tensor go,
                           int64_t stride, c10::optional<torch::Tensor> bn_ab_opt) {
  torch::Tensor bn_ab = bn_ab_opt.value_or(torch::Tensor());
  CHECK_GPU(x);
  auto xc = x.to(torch::kBFloat16).contiguous();
  auto g = go.to(torch::kBFloat16).contiguous();
  Conv2dDims cd;
  cd.N = (int)xc.size(0); cd.Cin = (int)xc.size(1);
  cd.H = (int)xc.size(2); cd.W = (int)xc.size(3);
  cd.stride = (int)stride;
  cd.Cout = (int)g.size(1);
  cd.OH = (int)g.size(2); cd.OW = (int)g.size(3);
  int K = cd.Cin * 9;
  int64_t M = (int64_t)cd.N * cd.OH * cd.OW;
  auto dw = torch::zeros({cd.Cout, (int64_t)K},
                         xc.options().dtype(torch::kFloat32));
  const bool fuse = bn_ab.defined() && bn_ab.numel() > 0;
  torch::Tensor ab;
  const float* abp = nullptr;
  if (fuse) {
    ab = bn_ab.to(torch::kFloat32).contiguous();
    TORCH_CHECK(ab.numel() == 2 * cd.Cin, "bn_ab must be [Cin,2]");
    abp = ab.data_ptr<float>();
  }
  int planes = ((cd.Cout + 31) / 32) * ((K + 31) / 32);
  int64_t target_chunks = std::max<int64_t>(1, 2048 / std::max(planes, 1));
  int64_t chunk = std::max<int64_t>(128, (M + target_chunks - 1) /
                                             target_chunks);
  chunk = ((chunk + 127) / 128) * 128;
  int64_t nchunks = (M + chunk - 1) / chunk;
  dim3 grid((cd.Cout + 31) / 32, (K + 31) / 32, (unsigned)nchunks);
  auto L = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<const __bf16*>(g.data_ptr()),
                       dw.data_ptr<float>(), cd, M, K, chunk, abp);
  };
  if (fuse) L(conv2d_wgrad_kernel<true>);
  else L(conv2d_wgrad_kernel<false>);
  return dw.view({cd.Cout, cd.Cin, 3, 3});
}
