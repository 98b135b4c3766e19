// K1 — implicit-GEMM Conv3d (3x3x3, pad 1, stride 1/2) on the matrix cores.
// The reference's flagship VBM workload runs 3D conv through stock PyTorch
// (SURVEY.md §2.9 K1); on gfx950 MIOpen degrades to naive fallbacks /
// im2col+GEMM for bf16 NCDHW (profiles/r01_vbm_baseline.md, ~3% of MFMA
// peak). Here: on-the-fly im2col staged through LDS feeding
// v_mfma_f32_16x16x32_bf16, fp32 accumulate.
//
// GEMM views (all NCDHW, w-fastest output index so gathers coalesce):
//   FWD  : C[M=N*OD*OH*OW, Cout] = patch(x)[M, Cin*27] @ W[Cout, Cin*27]^T
//   DGRAD: C[M=N*D*H*W,   Cin ] = scatter-gather(go)[M, Cout*27] @ Wf
//          (flipped kernel taps; stride handled by divisibility mask)
//   WGRAD: C[Cout, Cin*27] = go^T[Cout, M] @ patch(x)[M, Cin*27]
//          (split-K over M, fp32 atomics into the weight-grad buffer)
//
// Fragment layout (verified by the mfma_probe_* GPU tests): lane l holds
//   A[row = l&15][k = (l>>4)*8 + j]  (bf16x8 per lane)
//   B[k = (l>>4)*8 + j][col = l&15]
//   C/D col = l&15, row = (l>>4)*4 + r.
#include "common.h"

#include <hip/hip_bf16.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

struct ConvDims {
  int N, Cin, D, H, W;
  int Cout, OD, OH, OW;
  int stride;  // pad fixed at 1, kernel 3x3x3
};

// ---------------------------------------------------------------------------
// probe kernel: one wave computes C[16,16] = A[16,32] @ B[32,16]
// ---------------------------------------------------------------------------
__global__ void mfma_probe_kernel(const __bf16* __restrict__ A,
                                  const __bf16* __restrict__ B,
                                  float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
  const int row = lane & 15, kg = lane >> 4;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[row * 32 + kg * 8 + j];
    b[j] = B[(kg * 8 + j) * 16 + row];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[(kg * 4 + r) * 16 + (lane & 15)] = acc[r];
}

torch::Tensor mfma_probe_gemm(torch::Tensor A, torch::Tensor B) {
  CHECK_GPU(A);
  auto a = A.to(torch::kBFloat16).contiguous();
  auto b = B.to(torch::kBFloat16).contiguous();
  TORCH_CHECK(a.size(0) == 16 && a.size(1) == 32 && b.size(0) == 32 &&
              b.size(1) == 16, "probe expects A[16,32], B[32,16]");
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(a.data_ptr()),
                     reinterpret_cast<const __bf16*>(b.data_ptr()),
                     c.data_ptr<float>());
  return c;
}

// ---------------------------------------------------------------------------
// FWD / DGRAD: block = 256 threads (4 waves, 2x2), tile BM=64 x BN=64,
// K stepped 32 through LDS. Each wave: 32x32 C via 2x2 mfma fragments.
// ---------------------------------------------------------------------------
#define CBM 64
#define CBN 64
#define CBK 32
#define LDA_PAD 8  // bf16 elements of row padding (16B) — conflict-free b128 reads

// A-tile gather, FWD: m indexes output position, k = ci*27 + (kd,kh,kw)
__device__ inline __bf16 gather_fwd(const __bf16* __restrict__ x,
                                    const ConvDims& cd, int64_t m, int k) {
  const int ow = (int)(m % cd.OW);
  int64_t t = m / cd.OW;
  const int oh = (int)(t % cd.OH);
  t /= cd.OH;
  const int od = (int)(t % cd.OD);
  const int n = (int)(t / cd.OD);
  const int ci = k / 27;
  const int r = k - ci * 27;
  const int kd = r / 9, kh = (r / 3) % 3, kw = r % 3;
  const int id = od * cd.stride - 1 + kd;
  const int ih = oh * cd.stride - 1 + kh;
  const int iw = ow * cd.stride - 1 + kw;
  if ((unsigned)id >= (unsigned)cd.D || (unsigned)ih >= (unsigned)cd.H ||
      (unsigned)iw >= (unsigned)cd.W || ci >= cd.Cin)
    return (__bf16)0.f;
  return x[(((int64_t)n * cd.Cin + ci) * cd.D + id) * cd.H * cd.W +
           (int64_t)ih * cd.W + iw];
}

// A-tile gather, DGRAD: m indexes input position, k = co*27 + taps
__device__ inline __bf16 gather_dgrad(const __bf16* __restrict__ go,
                                      const ConvDims& cd, int64_t m, int k) {
  const int iw = (int)(m % cd.W);
  int64_t t = m / cd.W;
  const int ih = (int)(t % cd.H);
  t /= cd.H;
  const int id = (int)(t % cd.D);
  const int n = (int)(t / cd.D);
  const int co = k / 27;
  const int r = k - co * 27;
  const int kd = r / 9, kh = (r / 3) % 3, kw = r % 3;
  const int td = id + 1 - kd, th = ih + 1 - kh, tw = iw + 1 - kw;
  if (co >= cd.Cout) return (__bf16)0.f;
  if (td % cd.stride || th % cd.stride || tw % cd.stride) return (__bf16)0.f;
  const int od = td / cd.stride, oh = th / cd.stride, ow = tw / cd.stride;
  if ((unsigned)od >= (unsigned)cd.OD || (unsigned)oh >= (unsigned)cd.OH ||
      (unsigned)ow >= (unsigned)cd.OW)
    return (__bf16)0.f;
  return go[(((int64_t)n * cd.Cout + co) * cd.OD + od) * cd.OH * cd.OW +
            (int64_t)oh * cd.OW + ow];
}

// B-tile gather. FWD: B[k][col] = w[col][k] with w[Cout][Cin*27].
// DGRAD: B[k=(co,kd,kh,kw)][col=ci] = w[co][ci][26 - r] (flipped taps).
template <bool DGRAD>
__device__ inline __bf16 gather_w(const __bf16* __restrict__ w,
                                  const ConvDims& cd, int k, int col) {
  if (!DGRAD) {
    if (col >= cd.Cout || k >= cd.Cin * 27) return (__bf16)0.f;
    return w[(int64_t)col * (cd.Cin * 27) + k];
  }
  const int co = k / 27;
  const int r = k - co * 27;
  if (co >= cd.Cout || col >= cd.Cin) return (__bf16)0.f;
  // dgrad taps are NOT flipped here because gather_dgrad already maps
  // (id + 1 - kd): together they implement the transposed conv exactly.
  return w[((int64_t)co * cd.Cin + col) * 27 + r];
}

template <bool DGRAD, int STRIDE, bool FUSE_BN = false>
__global__ __launch_bounds__(256) void conv3d_igemm_kernel(
    const __bf16* __restrict__ Ain, const __bf16* __restrict__ w,
    __bf16* __restrict__ out, ConvDims cd, int64_t M, int Ncol, int K,
    const float* __restrict__ bn_ab = nullptr) {
  __shared__ __bf16 sA[CBM][CBK + LDA_PAD];
  __shared__ __bf16 sBT[CBN][CBK + LDA_PAD];  // [col][k]: lanes read 8
                                              // consecutive k => b128

  const int64_t bm = (int64_t)blockIdx.x * CBM;
  const int bn = blockIdx.y * CBN;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wm = (wave >> 1) * 32, wn = (wave & 1) * 32;

  // This thread stages ONE k column (kk_t) for 8 consecutive m positions.
  // Decode the m's once — the K loop then runs division-free (k-tap decode
  // uses compile-time-constant divisors the compiler strength-reduces).
  const int kk_t = tid >> 3;
  const int mbase = (tid * 8) & 63;
  const int SD = DGRAD ? cd.D : cd.OD;
  const int SH = DGRAD ? cd.H : cd.OH;
  const int SW = DGRAD ? cd.W : cd.OW;
  int pn[8], pd[8], ph[8], pw[8];
  {
    int64_t m = bm + mbase;
    int ww = (int)(m % SW);
    int64_t t = m / SW;
    int hh = (int)(t % SH);
    t /= SH;
    int dd = (int)(t % SD);
    int nn = (int)(t / SD);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      pn[j] = nn; pd[j] = dd; ph[j] = hh; pw[j] = ww;
      if (++ww == SW) { ww = 0; if (++hh == SH) { hh = 0;
          if (++dd == SD) { dd = 0; ++nn; } } }
    }
  }
  const bool m_ok = (bm + mbase + 7) < M;  // fast path: whole octet valid

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)cd.H * cd.W;
  const int64_t OHW = (int64_t)cd.OH * cd.OW;

  for (int k0 = 0; k0 < K; k0 += CBK) {
    const int k = k0 + kk_t;
    if (!DGRAD) {
      const int ci = k / 27;               // constant divisors: mul+shift
      const int r = k - ci * 27;
      const int kd = r / 9, kh = (r / 3) % 3, kw = r % 3;
      const bool k_ok = ci < cd.Cin;
      float a_c = 1.f, b_c = 0.f;
      if (FUSE_BN && k_ok) { a_c = bn_ab[ci * 2]; b_c = bn_ab[ci * 2 + 1]; }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __bf16 v = (__bf16)0.f;
        if (k_ok && (m_ok || (bm + mbase + j) < M)) {
          const int id = pd[j] * STRIDE - 1 + kd;
          const int ih = ph[j] * STRIDE - 1 + kh;
          const int iw = pw[j] * STRIDE - 1 + kw;
          if ((unsigned)id < (unsigned)cd.D && (unsigned)ih < (unsigned)cd.H &&
              (unsigned)iw < (unsigned)cd.W) {
            v = Ain[(((int64_t)pn[j] * cd.Cin + ci) * cd.D + id) * HW +
                    (int64_t)ih * cd.W + iw];
            if (FUSE_BN) v = (__bf16)fmaxf(a_c * (float)v + b_c, 0.f);
          }
        }
        sA[mbase + j][kk_t] = v;
      }
    } else {
      const int co = k / 27;
      const int r = k - co * 27;
      const int kd = r / 9, kh = (r / 3) % 3, kw = r % 3;
      const bool k_ok = co < cd.Cout;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __bf16 v = (__bf16)0.f;
        if (k_ok && (m_ok || (bm + mbase + j) < M)) {
          const int td = pd[j] + 1 - kd, th = ph[j] + 1 - kh,
                    tw = pw[j] + 1 - kw;
          if (STRIDE == 1 ||
              (!(td & (STRIDE - 1)) && !(th & (STRIDE - 1)) &&
               !(tw & (STRIDE - 1)))) {
            const int od = td / STRIDE, oh = th / STRIDE, ow = tw / STRIDE;
            if ((unsigned)od < (unsigned)cd.OD &&
                (unsigned)oh < (unsigned)cd.OH &&
                (unsigned)ow < (unsigned)cd.OW)
              v = Ain[(((int64_t)pn[j] * cd.Cout + co) * cd.OD + od) * OHW +
                      (int64_t)oh * cd.OW + ow];
          }
        }
        sA[mbase + j][kk_t] = v;
      }
    }
    // stage B (transposed): thread covers 8 consecutive k of one col so
    // the weight reads are contiguous and the LDS writes vectorize.
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      int idx = tid * 8 + e;
      int kk = idx & 31, col = idx >> 5;
      sBT[col][kk] = gather_w<DGRAD>(w, cd, k0 + kk, bn + col);
    }
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < CBK; ks += 32) {
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

  // epilogue: out[m][col] with col = channel — but the tensor layout is
  // [n][c][spatial], so out element = base(n) + col*spatialstride + m_sp.
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  const int64_t spatial = DGRAD ? (int64_t)cd.D * cd.H * cd.W
                                : (int64_t)cd.OD * cd.OH * cd.OW;
  const int nch = DGRAD ? cd.Cin : cd.Cout;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
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
  }
}

// ---------------------------------------------------------------------------
// WGRAD: C[Cout, Cin*27] = sum_m go[m,co] * patch(x)[m,k]; split-K over m
// with fp32 atomics. Block: 4 waves each owning a 16x16 (co x k) fragment
// pair; tile 32(co) x 32(k), K-chunk of positions per block.
// ---------------------------------------------------------------------------
#define WMB 128  // m positions staged per iteration (4 MFMA/wave/barrier)

template <bool FUSE_BN = false>
__global__ __launch_bounds__(256) void conv3d_wgrad_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ go,
    float* __restrict__ dw, ConvDims cd, int64_t M, int K, int64_t chunk,
    const float* __restrict__ bn_ab = nullptr) {
  __shared__ __bf16 sGoT[32][WMB + LDA_PAD];  // [co][m]: b128 frag reads
  __shared__ __bf16 sXT[32][WMB + LDA_PAD];   // [k][m]

  const int co0 = blockIdx.x * 32;
  const int k0 = blockIdx.y * 32;
  const int64_t m0 = (int64_t)blockIdx.z * chunk;
  const int64_t mEnd = min(m0 + chunk, M);
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wi = wave >> 1, wj = wave & 1;  // 2x2 over (co, k) 16x16 frags

  // this thread stages 4x4 m (four 32-m quads) for one co column AND one
  // k column — short 4-element runs keep lane addresses 8 B apart
  // (coalesced) while each barrier pair still feeds 4 MFMAs.
  const int mi0 = (tid & 7) * 4;
  const int ct = tid >> 3;          // co/k column (0..31)
  const int kq = k0 + ct;
  const int ci = kq / 27;
  const int rr = kq - ci * 27;
  const int kd = rr / 9, kh = (rr / 3) % 3, kw = rr % 3;
  const bool k_ok = ci < cd.Cin && kq < K;
  const bool c_ok = (co0 + ct) < cd.Cout;
  float a_c = 1.f, b_c = 0.f;
  if (FUSE_BN && k_ok) { a_c = bn_ab[ci * 2]; b_c = bn_ab[ci * 2 + 1]; }

  // incremental output-position decode for m = mb + mi0 (advances by 32)
  int nn, od, oh, ow;
  {
    int64_t m = m0 + mi0;
    ow = (int)(m % cd.OW);
    int64_t t = m / cd.OW;
    oh = (int)(t % cd.OH);
    t /= cd.OH;
    od = (int)(t % cd.OD);
    nn = (int)(t / cd.OD);
  }

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int64_t spatial = (int64_t)cd.OD * cd.OH * cd.OW;
  const int64_t HW = (int64_t)cd.H * cd.W;

  for (int64_t mb = m0; mb < mEnd; mb += WMB) {
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      int jn = nn, jd = od, jh = oh, jw = ow;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int mloc = q * 32 + mi0 + j;
        const int64_t m = mb + mloc;
        const bool ok = m < mEnd;
        __bf16 gv = (__bf16)0.f;
        if (ok && c_ok)
          gv = go[((int64_t)jn * cd.Cout + (co0 + ct)) * spatial +
                  ((int64_t)jd * cd.OH + jh) * cd.OW + jw];
        sGoT[ct][mloc] = gv;
        __bf16 xv = (__bf16)0.f;
        if (ok && k_ok) {
          const int id = jd * cd.stride - 1 + kd;
          const int ih = jh * cd.stride - 1 + kh;
          const int iw = jw * cd.stride - 1 + kw;
          if ((unsigned)id < (unsigned)cd.D &&
              (unsigned)ih < (unsigned)cd.H &&
              (unsigned)iw < (unsigned)cd.W) {
            xv = x[(((int64_t)jn * cd.Cin + ci) * cd.D + id) * HW +
                   (int64_t)ih * cd.W + iw];
            if (FUSE_BN) xv = (__bf16)fmaxf(a_c * (float)xv + b_c, 0.f);
          }
        }
        sXT[ct][mloc] = xv;
        if (++jw == cd.OW) { jw = 0; if (++jh == cd.OH) { jh = 0;
            if (++jd == cd.OD) { jd = 0; ++jn; } } }
      }
      // advance the walker by one 32-m quad
      ow += 32;
      while (ow >= cd.OW) {
        ow -= cd.OW;
        if (++oh == cd.OH) { oh = 0; if (++od == cd.OD) { od = 0; ++nn; } }
      }
    }
    __syncthreads();

    const int row = lane & 15, kg = lane >> 4;
#pragma unroll
    for (int ks = 0; ks < WMB; ks += 32) {
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

  const int ccol = lane & 15;          // k col within fragment
  const int crow0 = (lane >> 4) * 4;   // co row
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int co = co0 + wi * 16 + crow0 + r;
    int k = k0 + wj * 16 + ccol;
    if (co < cd.Cout && k < K)
      atomicAdd(&dw[(int64_t)co * K + k], acc[r]);
  }
}


// ---------------------------------------------------------------------------
// WGRAD, Cin=1 specialization (the VBM first layer): the split-K form
// gathers each x element once per tap column (27x traffic, 4-element
// scalar runs). Here a block owns a group of (n, od) slices: the single-
// channel x slab [3][OH+2][OW+2] stages ONCE per slice with 16B row
// vectors, go tiles [co][OHT*OW] stream through LDS, and the 27 taps form
// the MFMA B fragments read directly from the slab (taps = fragment rows,
// m = contraction). Per-block LDS reduction + one atomicAdd set per block
// keeps contention at blocks x Cout x 27.
// Requires: Cin==1, stride==1, OW % 32 == 0.
// ---------------------------------------------------------------------------
#define CI1_OHT 8

__global__ __launch_bounds__(256) void conv3d_wgrad_ci1_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ go,
    float* __restrict__ dw, ConvDims cd, int od_groups) {
  const int W2 = cd.W + 2;
  const int H2 = cd.H + 2;
  extern __shared__ __bf16 smem_ci1[];
  __bf16* sX = smem_ci1;                       // [3][H2][W2]
  __bf16* sGo = sX + 3 * H2 * W2;              // [32][CI1_OHT * OW + 8]
  const int GOL = CI1_OHT * cd.OW + 8;

  const int n = blockIdx.x;
  const int og = blockIdx.y;
  const int co0 = blockIdx.z * 32;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int row = lane & 15, kg = lane >> 4;

  const int64_t HW = (int64_t)cd.H * cd.W;
  const int64_t OHW = (int64_t)cd.OH * cd.OW;
  const int od_per = (cd.OD + od_groups - 1) / od_groups;
  const int od_lo = og * od_per;
  const int od_hi = min(cd.OD, od_lo + od_per);

  // acc[co frag][tap frag]: (16 co x 16 taps) x 2 x 2 per wave
  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int od = od_lo; od < od_hi; ++od) {
    // ---- stage the x slab for this od: rows (kd, ih) -------------------
    __syncthreads();
    for (int r = tid; r < 3 * H2; r += 256) {
      const int kd = r / H2;
      const int ihp = r % H2;            // padded row index: ih = ihp - 1
      const int id = od - 1 + kd;
      const int ih = ihp - 1;
      __bf16* dst = sX + ((int64_t)kd * H2 + ihp) * W2;
      if ((unsigned)id >= (unsigned)cd.D || (unsigned)ih >= (unsigned)cd.H) {
        for (int c = 0; c < W2; ++c) dst[c] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = x + ((int64_t)n * cd.D + id) * HW +
                          (int64_t)ih * cd.W;
      dst[0] = (__bf16)0.f;
      for (int v = 0; v + 8 <= cd.W; v += 8)
        *reinterpret_cast<bf16x8*>(dst + 1 + v) =
            *reinterpret_cast<const bf16x8*>(src + v);
      for (int c = (cd.W / 8) * 8; c < cd.W; ++c) dst[1 + c] = src[c];
      dst[1 + cd.W] = (__bf16)0.f;
    }

    for (int oh0 = 0; oh0 < cd.OH; oh0 += CI1_OHT) {
      // ---- stage go tile [co32][CI1_OHT * OW] --------------------------
      __syncthreads();
      for (int r = tid; r < 32 * CI1_OHT; r += 256) {
        const int ohl = r % CI1_OHT;
        const int co = r / CI1_OHT;
        __bf16* dst = sGo + (int64_t)co * GOL + ohl * cd.OW;
        const int oh = oh0 + ohl;
        if ((co0 + co) >= cd.Cout || oh >= cd.OH) {
          for (int c = 0; c < cd.OW; ++c) dst[c] = (__bf16)0.f;
          continue;
        }
        const __bf16* src = go + (((int64_t)n * cd.Cout + co0 + co) *
                                  cd.OD + od) * OHW + (int64_t)oh * cd.OW;
        for (int v = 0; v < cd.OW; v += 8)
          *reinterpret_cast<bf16x8*>(dst + v) =
              *reinterpret_cast<const bf16x8*>(src + v);
      }
      __syncthreads();

      // ---- MFMA: waves stride the m subchunks --------------------------
      const int nms = (CI1_OHT * cd.OW) / 32;
      for (int ms = wave; ms < nms; ms += 4) {
        const int m0 = ms * 32 + kg * 8;     // 8 consecutive m, no row wrap
        const int ohl = m0 / cd.OW;
        const int ow0 = m0 % cd.OW;
        bf16x8 afrag[2];
#pragma unroll
        for (int i = 0; i < 2; ++i) {
          const __bf16* src = sGo + (int64_t)(i * 16 + row) * GOL + m0;
#pragma unroll
          for (int j = 0; j < 8; ++j) afrag[i][j] = src[j];
        }
        bf16x8 bfrag[2];
#pragma unroll
        for (int t = 0; t < 2; ++t) {
          const int tap = t * 16 + row;      // 0..31; taps 27..31 padded
          const int kd = tap / 9, r9 = tap - kd * 9;
          const int kh = r9 / 3, kw = r9 % 3;
          if (tap < 27) {
            const __bf16* src = sX + ((int64_t)kd * H2 + (oh0 + ohl + kh)) *
                                W2 + ow0 + kw;
#pragma unroll
            for (int j = 0; j < 8; ++j) bfrag[t][j] = src[j];
          } else {
            bfrag[t] = bf16x8{};
          }
        }
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
          for (int t = 0; t < 2; ++t)
            acc[i][t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[i], bfrag[t], acc[i][t], 0, 0, 0);
      }
    }
  }

  // ---- cross-wave reduce through LDS, one atomic set per block ----------
  __syncthreads();
  float* red = reinterpret_cast<float*>(smem_ci1);  // reuse: 4*4*256 floats
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      const int fi = i * 2 + t;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        red[((int64_t)wave * 4 + fi) * 256 + lane * 4 + r] = acc[i][t][r];
    }
  __syncthreads();
  if (wave == 0) {
    // lane l, reg r of fragment fi: co = i*16 + (l>>4)*4+r? no — C layout:
    // row = (l>>4)*4+r (co), col = l&15 (tap)
#pragma unroll
    for (int fi = 0; fi < 4; ++fi) {
      const int i = fi >> 1, t = fi & 1;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = 0.f;
#pragma unroll
        for (int w = 0; w < 4; ++w)
          v += red[((int64_t)w * 4 + fi) * 256 + lane * 4 + r];
        const int co = co0 + i * 16 + (lane >> 4) * 4 + r;
        const int tap = t * 16 + (lane & 15);
        if (co < cd.Cout && tap < 27)
          atomicAdd(&dw[(int64_t)co * 27 + tap], v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Stride-2 DGRAD, parity-decomposed: the dense mask formulation wastes 7/8
// of the MFMA work (only taps with (id+1-kd) even contribute). Decompose
// dx by (id,ih,iw) mod 2 into 8 classes; each class is a DENSE implicit
// GEMM with K = Cout * (1 or 2)^3 taps — total FLOPs equal to the forward
// pass. blockIdx.z = class.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void conv3d_dgrad_s2_kernel(
    const __bf16* __restrict__ go, const __bf16* __restrict__ w,
    __bf16* __restrict__ dx, ConvDims cd) {
  __shared__ __bf16 sA[CBM][CBK + LDA_PAD];
  __shared__ __bf16 sBT[CBN][CBK + LDA_PAD];

  const int cls = blockIdx.z;
  const int a = (cls >> 2) & 1, b = (cls >> 1) & 1, c = cls & 1;
  const int Da = (cd.D - a + 1) >> 1;
  const int Hb = (cd.H - b + 1) >> 1;
  const int Wc = (cd.W - c + 1) >> 1;
  const int nd = a ? 2 : 1, nh = b ? 2 : 1, nw = c ? 2 : 1;
  const int l2w = c, l2h = b;              // log2 tap counts
  const int T = nd * nh * nw;
  const int l2T = a + b + c;
  const int K = cd.Cout << l2T;
  const int64_t M = (int64_t)cd.N * Da * Hb * Wc;

  const int64_t bm = (int64_t)blockIdx.x * CBM;
  if (bm >= M) return;
  const int bn = blockIdx.y * CBN;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wm = (wave >> 1) * 32, wn = (wave & 1) * 32;

  const int kk_t = tid >> 3;
  const int mbase = (tid * 8) & 63;
  int pn[8], pd[8], ph[8], pw[8];
  {
    int64_t m = bm + mbase;
    int ww = (int)(m % Wc);
    int64_t t = m / Wc;
    int hh = (int)(t % Hb);
    t /= Hb;
    int dd = (int)(t % Da);
    int nn = (int)(t / Da);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      pn[j] = nn; pd[j] = dd; ph[j] = hh; pw[j] = ww;
      if (++ww == Wc) { ww = 0; if (++hh == Hb) { hh = 0;
          if (++dd == Da) { dd = 0; ++nn; } } }
    }
  }
  const bool m_ok = (bm + mbase + 7) < M;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int64_t OHW = (int64_t)cd.OH * cd.OW;

  for (int k0 = 0; k0 < K; k0 += CBK) {
    {
      const int k = k0 + kk_t;
      const int co = k >> l2T;
      const int r = k & (T - 1);
      const int tw_i = r & (nw - 1);
      const int th_i = (r >> l2w) & (nh - 1);
      const int td_i = r >> (l2w + l2h);
      // tap: a==0 -> kd=1 (od=id'); a==1 -> kd=2*td_i (od=id'+1-td_i)
      const int dod = a ? (1 - td_i) : 0;   // od = id' + dod
      const int doh = b ? (1 - th_i) : 0;
      const int dow = c ? (1 - tw_i) : 0;
      const bool k_ok = co < cd.Cout;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __bf16 v = (__bf16)0.f;
        if (k_ok && (m_ok || (bm + mbase + j) < M)) {
          const int od = pd[j] + dod, oh = ph[j] + doh, ow = pw[j] + dow;
          if ((unsigned)od < (unsigned)cd.OD &&
              (unsigned)oh < (unsigned)cd.OH &&
              (unsigned)ow < (unsigned)cd.OW)
            v = go[(((int64_t)pn[j] * cd.Cout + co) * cd.OD + od) * OHW +
                   (int64_t)oh * cd.OW + ow];
        }
        sA[mbase + j][kk_t] = v;
      }
    }
    // B: w[co][ci=col][kd*9+kh*3+kw]
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      int idx = tid * 8 + e;
      int kk = idx & 31, col = idx >> 5;
      const int k = k0 + kk;
      const int co = k >> l2T;
      const int r = k & (T - 1);
      const int tw_i = r & (nw - 1);
      const int th_i = (r >> l2w) & (nh - 1);
      const int td_i = r >> (l2w + l2h);
      const int kd = a ? (td_i * 2) : 1;
      const int kh = b ? (th_i * 2) : 1;
      const int kw = c ? (tw_i * 2) : 1;
      __bf16 v = (__bf16)0.f;
      if (co < cd.Cout && (bn + col) < cd.Cin)
        v = w[((int64_t)co * cd.Cin + (bn + col)) * 27 + kd * 9 + kh * 3 +
              kw];
      sBT[col][kk] = v;
    }
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < CBK; ks += 32) {
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
  const int64_t HW = (int64_t)cd.H * cd.W;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int64_t m = bm + wm + i * 16 + crow0 + r;
        int ci = bn + wn + j * 16 + ccol;
        if (m < M && ci < cd.Cin) {
          const int iw = (int)(m % Wc);
          int64_t t = m / Wc;
          const int ih = (int)(t % Hb);
          t /= Hb;
          const int id = (int)(t % Da);
          const int n = (int)(t / Da);
          dx[(((int64_t)n * cd.Cin + ci) * cd.D + (2 * id + a)) * HW +
             (int64_t)(2 * ih + b) * cd.W + (2 * iw + c)] =
              (__bf16)(acc[i][j][r]);
        }
      }
    }
  }
}

static torch::Tensor conv3d_dgrad_s2(torch::Tensor g, torch::Tensor wc,
                                     torch::Tensor dx, ConvDims cd) {
  int Da = (cd.D + 1) >> 1, Hb = (cd.H + 1) >> 1, Wc = (cd.W + 1) >> 1;
  int64_t Mmax = (int64_t)cd.N * Da * Hb * Wc;
  dim3 grid((unsigned)((Mmax + CBM - 1) / CBM), (cd.Cin + CBN - 1) / CBN, 8);
  hipLaunchKernelGGL(conv3d_dgrad_s2_kernel, grid, dim3(256), 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(g.data_ptr()),
                     reinterpret_cast<const __bf16*>(wc.data_ptr()),
                     reinterpret_cast<__bf16*>(dx.data_ptr()), cd);
  return dx;
}

// ---------------------------------------------------------------------------
// WGRAD with tap reuse (stride 1 and 2): the split-K implicit-GEMM form
// re-reads x and go once PER (co,k)-plane (M*K element traffic ~ 27x the
// tensor). Here a block stages one x spatial slab [CT][3][H2][W2] (row-
// vectorized 16B loads) and one go tile [co][m] in LDS ONCE, then computes
// ALL 27 tap GEMMs from it: per 32-m sub-chunk the go fragment loads once
// and feeds 27 MFMAs whose x fragments read the same slab at (stride-
// scaled) tap-shifted offsets. STRIDE=1: waves 2x2 over (co32, ci32);
// STRIDE=2: waves 4x1 over (co64, ci16) with the strided window staged.
// Blocks grid-stride over spatial chunks; partials fold into dw by fp32
// atomics once at the end.
// ---------------------------------------------------------------------------
// SLICED: each z-block stores its partial (co, ci*27) tile into its own
// slice of dw[zstride][Cout][K] with plain stores; a cheap sum over z
// replaces the atomic fold (the r2 zstride sweep showed atomics cap the
// useful parallelism at ~512 blocks).
template <int OWT, int STRIDE, int CHUNK = 128, bool FUSE_BN = false,
          bool SLICED = false>
__global__ __launch_bounds__(256) void conv3d_wgrad_s1_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ go,
    float* __restrict__ dw, ConvDims cd, int64_t nchunks, int64_t zstride,
    const float* __restrict__ bn_ab = nullptr) {
  constexpr int OHT = CHUNK / OWT;
  constexpr int IW = STRIDE * OWT;
  constexpr int W2 = IW + (STRIDE == 1 ? 4 : 2);
  constexpr int H2 = STRIDE * (OHT - 1) + 3;
  constexpr int CT = STRIDE == 1 ? 32 : 16;     // ci tile
  constexpr int COT = STRIDE == 1 ? 32 : 64;    // co tile
  __shared__ __bf16 sX[CT][3][H2][W2];
  __shared__ __bf16 sGo[COT][CHUNK + LDA_PAD];

  const int co0 = blockIdx.x * COT;
  const int ci0 = blockIdx.y * CT;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  // fragment ownership: (co, ci) 16x16 per wave
  const int wi = (STRIDE == 1) ? (wave >> 1) : wave;
  const int wj = (STRIDE == 1) ? (wave & 1) : 0;
  const int row = lane & 15, kg = lane >> 4;

  const int wtiles = (cd.OW + OWT - 1) / OWT;
  const int htiles = (cd.OH + OHT - 1) / OHT;

  f32x4 acc[27];
#pragma unroll
  for (int t = 0; t < 27; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)cd.H * cd.W;
  const int64_t OHW = (int64_t)cd.OH * cd.OW;

  for (int64_t z = blockIdx.z; z < nchunks; z += zstride) {
    int64_t t = z;
    const int wt = (int)(t % wtiles);
    t /= wtiles;
    const int ht = (int)(t % htiles);
    t /= htiles;
    const int od = (int)(t % cd.OD);
    const int n = (int)(t / cd.OD);
    const int oh0 = ht * OHT, ow0 = wt * OWT;

    // ---- stage x slab row-wise (16B interior vectors, scalar halo) -----
    constexpr int NXROWS = CT * 3 * H2;
    const __bf16* xn = x + (int64_t)n * cd.Cin * cd.D * HW;
    for (int r = tid; r < NXROWS; r += 256) {
      const int hrow = r % H2;
      const int a = (r / H2) % 3;
      const int ci = r / (3 * H2);
      const int id = STRIDE * od - 1 + a;
      const int ih = STRIDE * oh0 - 1 + hrow;
      __bf16* dst = &sX[ci][a][hrow][0];
      const bool row_ok = (unsigned)id < (unsigned)cd.D &&
                          (unsigned)ih < (unsigned)cd.H &&
                          (ci0 + ci) < cd.Cin;
      if (!row_ok) {
#pragma unroll
        for (int col = 0; col < W2; ++col) dst[col] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = xn + ((int64_t)(ci0 + ci) * cd.D + id) * HW +
                          (int64_t)ih * cd.W;
      float a_c = 1.f, b_c = 0.f;
      if (FUSE_BN) {
        a_c = bn_ab[(ci0 + ci) * 2];
        b_c = bn_ab[(ci0 + ci) * 2 + 1];
      }
      auto tx = [&](__bf16 v) -> __bf16 {
        if (!FUSE_BN) return v;
        return (__bf16)fmaxf(a_c * (float)v + b_c, 0.f);
      };
      const int iw0 = STRIDE * ow0;
      dst[0] = (iw0 > 0) ? tx(src[iw0 - 1]) : (__bf16)0.f;
#pragma unroll
      for (int v = 0; v < IW / 8; ++v) {
        bf16x8 vec = *reinterpret_cast<const bf16x8*>(src + iw0 + v * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) dst[1 + v * 8 + j] = tx(vec[j]);
      }
#pragma unroll
      for (int e = 0; e < W2 - IW - 1; ++e) {
        const int iw = iw0 + IW + e;
        dst[1 + IW + e] = (iw < cd.W) ? tx(src[iw]) : (__bf16)0.f;
      }
    }
    // ---- stage go tile row-wise: [co][m over (OHT x OWT)] --------------
    const __bf16* gon = go + (int64_t)n * cd.Cout * cd.OD * OHW;
    for (int r = tid; r < COT * OHT; r += 256) {
      const int oh_off = r % OHT;
      const int co = r / OHT;
      __bf16* dst = &sGo[co][oh_off * OWT];
      const int oh = oh0 + oh_off;
      if ((co0 + co) >= cd.Cout || oh >= cd.OH) {
#pragma unroll
        for (int j = 0; j < OWT; ++j) dst[j] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = gon + ((int64_t)(co0 + co) * cd.OD + od) * OHW +
                          (int64_t)oh * cd.OW + ow0;
#pragma unroll
      for (int v = 0; v < OWT / 8; ++v)
        *reinterpret_cast<bf16x8*>(dst + v * 8) =
            *reinterpret_cast<const bf16x8*>(src + v * 8);
    }
    __syncthreads();

    // ---- m-subchunks x 27 taps -----------------------------------------
#pragma unroll 1
    for (int ms = 0; ms < CHUNK / 32; ++ms) {
      bf16x8 afrag;   // go[co16][m32]
#pragma unroll
      for (int j = 0; j < 8; ++j)
        afrag[j] = sGo[wi * 16 + row][ms * 32 + kg * 8 + j];
      const int mbase = ms * 32 + kg * 8;
      const int oh_off = mbase / OWT;
      const int ow_off = mbase % OWT;
#pragma unroll
      for (int kd = 0; kd < 3; ++kd) {
#pragma unroll
        for (int kh = 0; kh < 3; ++kh) {
#pragma unroll
          for (int kw = 0; kw < 3; ++kw) {
            bf16x8 bfrag;
            const __bf16* src = &sX[wj * 16 + row][kd]
                                   [STRIDE * oh_off + kh]
                                   [STRIDE * ow_off + kw];
#pragma unroll
            for (int j = 0; j < 8; ++j) bfrag[j] = src[STRIDE * j];
            acc[(kd * 3 + kh) * 3 + kw] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afrag, bfrag, acc[(kd * 3 + kh) * 3 + kw], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  // ---- fold partials into dw[co][ci*27 + tap] -------------------------
  const int K = cd.Cin * 27;
  float* out = SLICED ? dw + (int64_t)blockIdx.z * cd.Cout * K : dw;
  const int ccol = lane & 15;          // ci col within fragment
  const int crow0 = (lane >> 4) * 4;   // co row
#pragma unroll 1
  for (int tp = 0; tp < 27; ++tp) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int co = co0 + wi * 16 + crow0 + r;
      const int ci = ci0 + wj * 16 + ccol;
      if (co < cd.Cout && ci < cd.Cin) {
        if (SLICED)
          out[(int64_t)co * K + ci * 27 + tp] = acc[tp][r];
        else
          atomicAdd(&out[(int64_t)co * K + ci * 27 + tp], acc[tp][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Double-buffered wgrad (experimental; routed via variant=1, stride 1):
// stage the (x slab, go tile) of chunk z+zstride while chunk z's
// 4 m-subchunks x 27 tap-MFMAs run — one barrier per chunk instead of
// two, attacking the 67% SQ_WAIT_ANY the profiles show. LDS doubles to
// ~114 KB at OWT=32 (1 block/CU): whether intra-block overlap beats the
// lost co-residency is the round-2 measurement.
// ---------------------------------------------------------------------------
template <int OWT, int STRIDE, int CHUNK = 128>
__global__ __launch_bounds__(256) void conv3d_wgrad_s1_db_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ go,
    float* __restrict__ dw, ConvDims cd, int64_t nchunks, int64_t zstride) {
  constexpr int OHT = CHUNK / OWT;
  constexpr int IW = STRIDE * OWT;
  constexpr int W2 = IW + (STRIDE == 1 ? 4 : 2);
  constexpr int H2 = STRIDE * (OHT - 1) + 3;
  constexpr int CT = STRIDE == 1 ? 32 : 16;
  constexpr int COT = STRIDE == 1 ? 32 : 64;
  __shared__ __bf16 sX[2][CT][3][H2][W2];
  __shared__ __bf16 sGo[2][COT][CHUNK + LDA_PAD];

  const int co0 = blockIdx.x * COT;
  const int ci0 = blockIdx.y * CT;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wi = (STRIDE == 1) ? (wave >> 1) : wave;
  const int wj = (STRIDE == 1) ? (wave & 1) : 0;
  const int row = lane & 15, kg = lane >> 4;

  const int wtiles = (cd.OW + OWT - 1) / OWT;
  const int htiles = (cd.OH + OHT - 1) / OHT;

  f32x4 acc[27];
#pragma unroll
  for (int t = 0; t < 27; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)cd.H * cd.W;
  const int64_t OHW = (int64_t)cd.OH * cd.OW;

  auto stage = [&](int64_t z, int buf) {
    int64_t t = z;
    const int wt = (int)(t % wtiles);
    t /= wtiles;
    const int ht = (int)(t % htiles);
    t /= htiles;
    const int od = (int)(t % cd.OD);
    const int n = (int)(t / cd.OD);
    const int oh0 = ht * OHT, ow0 = wt * OWT;

    constexpr int NXROWS = CT * 3 * H2;
    const __bf16* xn = x + (int64_t)n * cd.Cin * cd.D * HW;
    for (int r = tid; r < NXROWS; r += 256) {
      const int hrow = r % H2;
      const int a = (r / H2) % 3;
      const int ci = r / (3 * H2);
      const int id = STRIDE * od - 1 + a;
      const int ih = STRIDE * oh0 - 1 + hrow;
      __bf16* dst = &sX[buf][ci][a][hrow][0];
      const bool row_ok = (unsigned)id < (unsigned)cd.D &&
                          (unsigned)ih < (unsigned)cd.H &&
                          (ci0 + ci) < cd.Cin;
      if (!row_ok) {
#pragma unroll
        for (int col = 0; col < W2; ++col) dst[col] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = xn + ((int64_t)(ci0 + ci) * cd.D + id) * HW +
                          (int64_t)ih * cd.W;
      const int iw0 = STRIDE * ow0;
      dst[0] = (iw0 > 0) ? src[iw0 - 1] : (__bf16)0.f;
#pragma unroll
      for (int v = 0; v < IW / 8; ++v) {
        bf16x8 vec = *reinterpret_cast<const bf16x8*>(src + iw0 + v * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) dst[1 + v * 8 + j] = vec[j];
      }
#pragma unroll
      for (int e = 0; e < W2 - IW - 1; ++e) {
        const int iw = iw0 + IW + e;
        dst[1 + IW + e] = (iw < cd.W) ? src[iw] : (__bf16)0.f;
      }
    }
    const __bf16* gon = go + (int64_t)n * cd.Cout * cd.OD * OHW;
    for (int r = tid; r < COT * OHT; r += 256) {
      const int oh_off = r % OHT;
      const int co = r / OHT;
      __bf16* dst = &sGo[buf][co][oh_off * OWT];
      const int oh = oh0 + oh_off;
      if ((co0 + co) >= cd.Cout || oh >= cd.OH) {
#pragma unroll
        for (int j = 0; j < OWT; ++j) dst[j] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = gon + ((int64_t)(co0 + co) * cd.OD + od) * OHW +
                          (int64_t)oh * cd.OW + ow0;
#pragma unroll
      for (int v = 0; v < OWT / 8; ++v)
        *reinterpret_cast<bf16x8*>(dst + v * 8) =
            *reinterpret_cast<const bf16x8*>(src + v * 8);
    }
  };

  auto compute = [&](int buf) {
#pragma unroll 1
    for (int ms = 0; ms < CHUNK / 32; ++ms) {
      bf16x8 afrag;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        afrag[j] = sGo[buf][wi * 16 + row][ms * 32 + kg * 8 + j];
      const int mbase = ms * 32 + kg * 8;
      const int oh_off = mbase / OWT;
      const int ow_off = mbase % OWT;
#pragma unroll
      for (int kd = 0; kd < 3; ++kd) {
#pragma unroll
        for (int kh = 0; kh < 3; ++kh) {
#pragma unroll
          for (int kw = 0; kw < 3; ++kw) {
            bf16x8 bfrag;
            const __bf16* src = &sX[buf][wj * 16 + row][kd]
                                   [STRIDE * oh_off + kh]
                                   [STRIDE * ow_off + kw];
#pragma unroll
            for (int j = 0; j < 8; ++j) bfrag[j] = src[STRIDE * j];
            acc[(kd * 3 + kh) * 3 + kw] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afrag, bfrag, acc[(kd * 3 + kh) * 3 + kw], 0, 0, 0);
          }
        }
      }
    }
  };

  int buf = 0;
  if (blockIdx.z < nchunks) stage(blockIdx.z, 0);
  __syncthreads();
  for (int64_t z = blockIdx.z; z < nchunks; z += zstride) {
    if (z + zstride < nchunks) stage(z + zstride, buf ^ 1);
    compute(buf);
    __syncthreads();  // next chunk staged AND this buffer's reads done
    buf ^= 1;
  }

  const int K = cd.Cin * 27;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll 1
  for (int tp = 0; tp < 27; ++tp) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int co = co0 + wi * 16 + crow0 + r;
      const int ci = ci0 + wj * 16 + ccol;
      if (co < cd.Cout && ci < cd.Cin)
        atomicAdd(&dw[(int64_t)co * K + ci * 27 + tp], acc[tp][r]);
    }
  }
}

// bias grad + (optionally) any channelwise sums: dB[co] = sum over m of go
__global__ void channel_sum_kernel(const __bf16* __restrict__ go,
                                   float* __restrict__ db, int N, int C,
                                   int64_t spatial) {
  // one block per channel; waves stride the spatial x batch space
  const int c = blockIdx.x;
  float s = 0.f;
  for (int64_t i = threadIdx.x; i < (int64_t)N * spatial; i += blockDim.x) {
    int64_t n = i / spatial, sp = i % spatial;
    s += (float)go[((int64_t)n * C + c) * spatial + sp];
  }
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) s += __shfl_down(s, off);
  __shared__ float partial[4];
  if ((threadIdx.x & 63) == 0) partial[threadIdx.x >> 6] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
    for (int i = 0; i < (int)(blockDim.x >> 6); ++i) t += partial[i];
    db[c] = t;
  }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------
static ConvDims make_dims(const torch::Tensor& x, const torch::Tensor& w,
                          int stride) {
  ConvDims cd;
  cd.N = (int)x.size(0); cd.Cin = (int)x.size(1);
  cd.D = (int)x.size(2); cd.H = (int)x.size(3); cd.W = (int)x.size(4);
  cd.Cout = (int)w.size(0);
  cd.stride = stride;
  cd.OD = (cd.D + 2 - 3) / stride + 1;
  cd.OH = (cd.H + 2 - 3) / stride + 1;
  cd.OW = (cd.W + 2 - 3) / stride + 1;
  return cd;
}

torch::Tensor conv3d_fwd(torch::Tensor x, torch::Tensor w, int64_t stride,
                         c10::optional<torch::Tensor> bn_ab_opt) {
  torch::Tensor bn_ab = bn_ab_opt.value_or(torch::Tensor());
  CHECK_GPU(x);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "conv3d_fwd wants bf16");
  auto xc = x.contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  auto cd = make_dims(xc, wc, (int)stride);
  TORCH_CHECK(wc.size(2) == 3 && wc.size(3) == 3 && wc.size(4) == 3 &&
              wc.size(1) == cd.Cin, "3x3x3 kernels only");
  const bool fuse = bn_ab.defined() && bn_ab.numel() > 0;
  torch::Tensor ab;
  const float* abp = nullptr;
  if (fuse) {
    ab = bn_ab.to(torch::kFloat32).contiguous();
    TORCH_CHECK(ab.numel() == 2 * cd.Cin, "bn_ab must be [Cin,2]");
    abp = ab.data_ptr<float>();
  }
  auto out = torch::empty({cd.N, cd.Cout, cd.OD, cd.OH, cd.OW}, xc.options());
  int64_t M = (int64_t)cd.N * cd.OD * cd.OH * cd.OW;
  int K = cd.Cin * 27;
  dim3 grid((unsigned)((M + CBM - 1) / CBM), (cd.Cout + CBN - 1) / CBN);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<const __bf16*>(wc.data_ptr()),
                       reinterpret_cast<__bf16*>(out.data_ptr()), cd, M,
                       cd.Cout, K, abp);
  };
  TORCH_CHECK(stride == 1 || stride == 2, "stride must be 1 or 2");
  if (fuse) {
    if (stride == 1) launch(conv3d_igemm_kernel<false, 1, true>);
    else launch(conv3d_igemm_kernel<false, 2, true>);
  } else if (stride == 1) {
    launch(conv3d_igemm_kernel<false, 1>);
  } else {
    launch(conv3d_igemm_kernel<false, 2>);
  }
  return out;
}

torch::Tensor conv3d_dgrad(torch::Tensor go, torch::Tensor w,
                           std::vector<int64_t> in_shape, int64_t stride) {
  CHECK_GPU(go);
  auto g = go.to(torch::kBFloat16).contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  auto dx = torch::empty(in_shape, g.options());
  ConvDims cd;
  cd.N = (int)in_shape[0]; cd.Cin = (int)in_shape[1];
  cd.D = (int)in_shape[2]; cd.H = (int)in_shape[3]; cd.W = (int)in_shape[4];
  cd.Cout = (int)wc.size(0);
  cd.stride = (int)stride;
  cd.OD = (int)g.size(2); cd.OH = (int)g.size(3); cd.OW = (int)g.size(4);
  int64_t M = (int64_t)cd.N * cd.D * cd.H * cd.W;
  int K = cd.Cout * 27;
  TORCH_CHECK(stride == 1 || stride == 2, "stride must be 1 or 2");
  if (stride == 2) return conv3d_dgrad_s2(g, wc, dx, cd);
  dim3 grid((unsigned)((M + CBM - 1) / CBM), (cd.Cin + CBN - 1) / CBN);
  hipLaunchKernelGGL((conv3d_igemm_kernel<true, 1>), grid, dim3(256), 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(g.data_ptr()),
                     reinterpret_cast<const __bf16*>(wc.data_ptr()),
                     reinterpret_cast<__bf16*>(dx.data_ptr()), cd, M, cd.Cin,
                     K, (const float*)nullptr);
  return dx;
}

torch::Tensor conv3d_wgrad(torch::Tensor x, torch::Tensor go,
                           int64_t stride, int64_t variant,
                           c10::optional<torch::Tensor> bn_ab_opt) {
  torch::Tensor bn_ab = bn_ab_opt.value_or(torch::Tensor());
  CHECK_GPU(x);
  auto xc = x.to(torch::kBFloat16).contiguous();
  auto g = go.to(torch::kBFloat16).contiguous();
  ConvDims cd;
  cd.N = (int)xc.size(0); cd.Cin = (int)xc.size(1);
  cd.D = (int)xc.size(2); cd.H = (int)xc.size(3); cd.W = (int)xc.size(4);
  cd.Cout = (int)g.size(1);
  cd.stride = (int)stride;
  cd.OD = (int)g.size(2); cd.OH = (int)g.size(3); cd.OW = (int)g.size(4);
  int K = cd.Cin * 27;
  int64_t M = (int64_t)cd.N * cd.OD * cd.OH * cd.OW;
  auto dw = torch::zeros({cd.Cout, (int64_t)K},
                         xc.options().dtype(torch::kFloat32));
  const bool fuse = bn_ab.defined() && bn_ab.numel() > 0;
  torch::Tensor ab;
  const float* abp = nullptr;
  if (fuse) {
    TORCH_CHECK(variant == 0, "fused BN wgrad: default variant only");
    ab = bn_ab.to(torch::kFloat32).contiguous();
    TORCH_CHECK(ab.numel() == 2 * cd.Cin, "bn_ab must be [Cin,2]");
    abp = ab.data_ptr<float>();
  }

  if (cd.Cin == 1 && stride == 1 && cd.OW % 32 == 0 && !fuse) {
    // Cin=1 first-layer specialization (slab tap-reuse, see kernel doc)
    const int od_groups = std::max(1, std::min(cd.OD, 512 / cd.N));
    const int W2 = cd.W + 2, H2 = cd.H + 2;
    const int GOL = CI1_OHT * cd.OW + 8;
    size_t lds = (size_t)(3 * H2 * W2 + 32 * GOL) * sizeof(__bf16);
    if (lds <= 64 * 1024) {
      dim3 grid(cd.N, od_groups, (cd.Cout + 31) / 32);
      hipLaunchKernelGGL(conv3d_wgrad_ci1_kernel, grid, dim3(256), lds,
                         current_stream(),
                         reinterpret_cast<const __bf16*>(xc.data_ptr()),
                         reinterpret_cast<const __bf16*>(g.data_ptr()),
                         dw.data_ptr<float>(), cd, od_groups);
      return dw.view({cd.Cout, cd.Cin, 3, 3, 3});
    }
  }
  if ((cd.OW % 8) == 0 && cd.Cin >= 16 && cd.OH * cd.OW >= 64 &&
      (stride == 1 || stride == 2)) {
    // tap-reuse path (stride-templated)
    int OWT = cd.OW % 32 == 0 ? 32 : (cd.OW % 16 == 0 ? 16 : 8);
    int chunk = 128;
    if (cd.OH * cd.OW < 128) {
      chunk = 64;
      OWT = 8;
    }
    int wtiles = (cd.OW + OWT - 1) / OWT;
    int OHT = chunk / OWT;
    int htiles = (cd.OH + OHT - 1) / OHT;
    int COT = stride == 1 ? 32 : 64, CT = stride == 1 ? 32 : 16;
    int co_t = (cd.Cout + COT - 1) / COT, ci_t = (cd.Cin + CT - 1) / CT;
    int64_t nchunks = (int64_t)cd.N * cd.OD * htiles * wtiles;
    // blocks-in-flight target: swept on MI355X (r2) — 512 won (41.0
    // ms/step vs 42.1 at the old 768; >=1024 loses to atomic-fold
    // contention, <=256 underfills the chip). COINN_WGRAD_Z overrides.
    static const int zbase = []() {
      const char* e = getenv("COINN_WGRAD_Z");
      return e ? atoi(e) : 512;
    }();
    int64_t zstride = std::max<int64_t>(
        1, std::min<int64_t>(nchunks, zbase / std::max(co_t * ci_t, 1)));
    // sliced partial buffers pay when z-parallelism is high relative to
    // the (co, ci) tile count (atomic-fold contention regime)
    const bool sliced = (variant == 0) && zstride >= 32 &&
                        (int64_t)zstride * cd.Cout * K * 4 <=
                            (int64_t)512 * 1024 * 1024;
    torch::Tensor part;
    float* outp = dw.data_ptr<float>();
    if (sliced) {
      part = torch::empty({(int64_t)zstride, (int64_t)cd.Cout, (int64_t)K},
                          xc.options().dtype(torch::kFloat32));
      outp = part.data_ptr<float>();
    }
    dim3 grid(co_t, ci_t, (unsigned)zstride);
    auto L = [&](auto kern) {
      hipLaunchKernelGGL(kern, grid, dim3(256), 0, current_stream(),
                         reinterpret_cast<const __bf16*>(xc.data_ptr()),
                         reinterpret_cast<const __bf16*>(g.data_ptr()),
                         outp, cd, nchunks, zstride, abp);
    };
    if (sliced) {
      if (fuse) {
        if (stride == 1) {
          if (chunk == 64) L(conv3d_wgrad_s1_kernel<8, 1, 64, true, true>);
          else if (OWT == 32)
            L(conv3d_wgrad_s1_kernel<32, 1, 128, true, true>);
          else if (OWT == 16)
            L(conv3d_wgrad_s1_kernel<16, 1, 128, true, true>);
          else L(conv3d_wgrad_s1_kernel<8, 1, 128, true, true>);
        } else {
          if (chunk == 64) L(conv3d_wgrad_s1_kernel<8, 2, 64, true, true>);
          else if (OWT == 32)
            L(conv3d_wgrad_s1_kernel<32, 2, 128, true, true>);
          else if (OWT == 16)
            L(conv3d_wgrad_s1_kernel<16, 2, 128, true, true>);
          else L(conv3d_wgrad_s1_kernel<8, 2, 128, true, true>);
        }
      } else if (stride == 1) {
        if (chunk == 64) L(conv3d_wgrad_s1_kernel<8, 1, 64, false, true>);
        else if (OWT == 32)
          L(conv3d_wgrad_s1_kernel<32, 1, 128, false, true>);
        else if (OWT == 16)
          L(conv3d_wgrad_s1_kernel<16, 1, 128, false, true>);
        else L(conv3d_wgrad_s1_kernel<8, 1, 128, false, true>);
      } else {
        if (chunk == 64) L(conv3d_wgrad_s1_kernel<8, 2, 64, false, true>);
        else if (OWT == 32)
          L(conv3d_wgrad_s1_kernel<32, 2, 128, false, true>);
        else if (OWT == 16)
          L(conv3d_wgrad_s1_kernel<16, 2, 128, false, true>);
        else L(conv3d_wgrad_s1_kernel<8, 2, 128, false, true>);
      }
      auto dwsum = part.sum(0);
      return dwsum.view({cd.Cout, cd.Cin, 3, 3, 3});
    }
    if (fuse) {
      if (stride == 1) {
        if (chunk == 64) L(conv3d_wgrad_s1_kernel<8, 1, 64, true>);
        else if (OWT == 32) L(conv3d_wgrad_s1_kernel<32, 1, 128, true>);
        else if (OWT == 16) L(conv3d_wgrad_s1_kernel<16, 1, 128, true>);
        else L(conv3d_wgrad_s1_kernel<8, 1, 128, true>);
      } else {
        if (chunk == 64) L(conv3d_wgrad_s1_kernel<8, 2, 64, true>);
        else if (OWT == 32) L(conv3d_wgrad_s1_kernel<32, 2, 128, true>);
        else if (OWT == 16) L(conv3d_wgrad_s1_kernel<16, 2, 128, true>);
        else L(conv3d_wgrad_s1_kernel<8, 2, 128, true>);
      }
      return dw.view({cd.Cout, cd.Cin, 3, 3, 3});
    }
    auto LDB = [&](auto kern) {
      hipLaunchKernelGGL(kern, grid, dim3(256), 0, current_stream(),
                         reinterpret_cast<const __bf16*>(xc.data_ptr()),
                         reinterpret_cast<const __bf16*>(g.data_ptr()),
                         dw.data_ptr<float>(), cd, nchunks, zstride);
    };
    if (stride == 1 && variant == 1) {
      // experimental double-buffered instances
      if (chunk == 64) LDB(conv3d_wgrad_s1_db_kernel<8, 1, 64>);
      else if (OWT == 32) LDB(conv3d_wgrad_s1_db_kernel<32, 1>);
      else if (OWT == 16) LDB(conv3d_wgrad_s1_db_kernel<16, 1>);
      else LDB(conv3d_wgrad_s1_db_kernel<8, 1>);
    } else if (stride == 1) {
      if (chunk == 64) L(conv3d_wgrad_s1_kernel<8, 1, 64>);
      else if (OWT == 32) L(conv3d_wgrad_s1_kernel<32, 1>);
      else if (OWT == 16) L(conv3d_wgrad_s1_kernel<16, 1>);
      else L(conv3d_wgrad_s1_kernel<8, 1>);
    } else {
      if (chunk == 64) L(conv3d_wgrad_s1_kernel<8, 2, 64>);
      else if (OWT == 32) L(conv3d_wgrad_s1_kernel<32, 2>);
      else if (OWT == 16) L(conv3d_wgrad_s1_kernel<16, 2>);
      else L(conv3d_wgrad_s1_kernel<8, 2>);
    }
    return dw.view({cd.Cout, cd.Cin, 3, 3, 3});
  }

  // fallback: split-K implicit GEMM (any shape / stride)
  int planes = ((cd.Cout + 31) / 32) * ((K + 31) / 32);
  int64_t target_chunks = std::max<int64_t>(1, 2048 / std::max(planes, 1));
  int64_t chunk = std::max<int64_t>(128, (M + target_chunks - 1) /
                                              target_chunks);
  chunk = ((chunk + 127) / 128) * 128;
  int64_t nchunks = (M + chunk - 1) / chunk;
  dim3 grid((cd.Cout + 31) / 32, (K + 31) / 32, (unsigned)nchunks);
  auto LSK = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<const __bf16*>(g.data_ptr()),
                       dw.data_ptr<float>(), cd, M, K, chunk, abp);
  };
  if (fuse) LSK(conv3d_wgrad_kernel<true>);
  else LSK(conv3d_wgrad_kernel<false>);
  return dw.view({cd.Cout, cd.Cin, 3, 3, 3});
}

torch::Tensor channel_sum(torch::Tensor go) {
  CHECK_GPU(go);
  auto g = go.to(torch::kBFloat16).contiguous();
  int N = (int)g.size(0), C = (int)g.size(1);
  int64_t spatial = g.numel() / ((int64_t)N * C);
  auto out = torch::empty({C}, g.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(channel_sum_kernel, dim3(C), dim3(256), 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(g.data_ptr()),
                     out.data_ptr<float>(), N, C, spatial);
  return out;
}
