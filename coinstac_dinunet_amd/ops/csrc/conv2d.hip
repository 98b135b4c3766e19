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
  int stride;  // pad = KS/2; kernel KS x KS (KS = 3, or 7 for the stem)
};

#define CBM2 64
#define CBN2 64
#define CBK2 32
#define LDA_PAD2 8

template <bool DGRAD, int KS = 3>
__device__ inline __bf16 gather2_w(const __bf16* __restrict__ w,
                                   const Conv2dDims& cd, int k, int col) {
  constexpr int KK = KS * KS;
  if (!DGRAD) {
    if (col >= cd.Cout || k >= cd.Cin * KK) return (__bf16)0.f;
    return w[(int64_t)col * (cd.Cin * KK) + k];
  }
  const int co = k / KK;
  const int r = k - co * KK;
  if (co >= cd.Cout || col >= cd.Cin) return (__bf16)0.f;
  // taps un-flipped here; the (ih + 1 - kh) mapping in the A gather
  // implements the transposed conv (same convention as conv3d.hip)
  return w[((int64_t)co * cd.Cin + col) * KK + r];
}

template <bool DGRAD, int STRIDE, bool FUSE_BN = false, int KS = 3>
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
    constexpr int KK = KS * KS;
    constexpr int PAD = KS / 2;
    if (!DGRAD) {
      const int ci = k / KK;
      const int r = k - ci * KK;
      const int kh = r / KS, kw = r % KS;
      const bool k_ok = ci < cd.Cin;
      float a_c = 1.f, b_c = 0.f;
      if (FUSE_BN && k_ok) { a_c = bn_ab[ci * 2]; b_c = bn_ab[ci * 2 + 1]; }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __bf16 v = (__bf16)0.f;
        if (k_ok && (m_ok || (bm + mbase + j) < M)) {
          const int ih = ph[j] * STRIDE - PAD + kh;
          const int iw = pw[j] * STRIDE - PAD + kw;
          if ((unsigned)ih < (unsigned)cd.H && (unsigned)iw < (unsigned)cd.W) {
            v = Ain[((int64_t)pn[j] * cd.Cin + ci) * HW +
                    (int64_t)ih * cd.W + iw];
            if (FUSE_BN) v = (__bf16)fmaxf(a_c * (float)v + b_c, 0.f);
          }
        }
        sA[mbase + j][kk_t] = v;
      }
    } else {
      const int co = k / KK;
      const int r = k - co * KK;
      const int kh = r / KS, kw = r % KS;
      const bool k_ok = co < cd.Cout;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __bf16 v = (__bf16)0.f;
        if (k_ok && (m_ok || (bm + mbase + j) < M)) {
          const int th = ph[j] + PAD - kh, tw = pw[j] + PAD - kw;
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
      sBT[col][kk] = gather2_w<DGRAD, KS>(w, cd, k0 + kk, bn + col);
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
// Stride-2 DGRAD, parity-decomposed (2D port of conv3d_dgrad_s2_kernel):
// the masked formulation wastes 3/4 of the MFMA work; decompose dx by
// (ih, iw) mod 2 into 4 classes, each a DENSE implicit GEMM with
// K = Cout * (1 or 2)^2 taps. blockIdx.z = class.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void conv2d_dgrad_s2_kernel(
    const __bf16* __restrict__ go, const __bf16* __restrict__ w,
    __bf16* __restrict__ dx, Conv2dDims cd) {
  __shared__ __bf16 sA[CBM2][CBK2 + LDA_PAD2];
  __shared__ __bf16 sBT[CBN2][CBK2 + LDA_PAD2];

  const int cls = blockIdx.z;
  const int b = (cls >> 1) & 1, c = cls & 1;
  const int Hb = (cd.H - b + 1) >> 1;
  const int Wc = (cd.W - c + 1) >> 1;
  const int nh = b ? 2 : 1, nw = c ? 2 : 1;
  const int l2w = c;
  const int T = nh * nw;
  const int l2T = b + c;
  const int K = cd.Cout << l2T;
  const int64_t M = (int64_t)cd.N * Hb * Wc;

  const int64_t bm = (int64_t)blockIdx.x * CBM2;
  if (bm >= M) return;
  const int bn = blockIdx.y * CBN2;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wm = (wave >> 1) * 32, wn = (wave & 1) * 32;

  const int kk_t = tid >> 3;
  const int mbase = (tid * 8) & 63;
  int pn[8], ph[8], pw[8];
  {
    int64_t m = bm + mbase;
    int ww = (int)(m % Wc);
    int64_t t = m / Wc;
    int hh = (int)(t % Hb);
    int nn = (int)(t / Hb);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      pn[j] = nn; ph[j] = hh; pw[j] = ww;
      if (++ww == Wc) { ww = 0; if (++hh == Hb) { hh = 0; ++nn; } }
    }
  }
  const bool m_ok = (bm + mbase + 7) < M;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int64_t OHW = (int64_t)cd.OH * cd.OW;

  for (int k0 = 0; k0 < K; k0 += CBK2) {
    {
      const int k = k0 + kk_t;
      const int co = k >> l2T;
      const int r = k & (T - 1);
      const int tw_i = r & (nw - 1);
      const int th_i = r >> l2w;
      const int doh = b ? (1 - th_i) : 0;  // oh = ih' + doh
      const int dow = c ? (1 - tw_i) : 0;
      const bool k_ok = co < cd.Cout;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __bf16 v = (__bf16)0.f;
        if (k_ok && (m_ok || (bm + mbase + j) < M)) {
          const int oh = ph[j] + doh, ow = pw[j] + dow;
          if ((unsigned)oh < (unsigned)cd.OH &&
              (unsigned)ow < (unsigned)cd.OW)
            v = go[((int64_t)pn[j] * cd.Cout + co) * OHW +
                   (int64_t)oh * cd.OW + ow];
        }
        sA[mbase + j][kk_t] = v;
      }
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      int idx = tid * 8 + e;
      int kk = idx & 31, col = idx >> 5;
      const int k = k0 + kk;
      const int co = k >> l2T;
      const int r = k & (T - 1);
      const int tw_i = r & (nw - 1);
      const int th_i = r >> l2w;
      const int kh = b ? (th_i * 2) : 1;
      const int kw = c ? (tw_i * 2) : 1;
      __bf16 v = (__bf16)0.f;
      if (co < cd.Cout && (bn + col) < cd.Cin)
        v = w[((int64_t)co * cd.Cin + (bn + col)) * 9 + kh * 3 + kw];
      sBT[col][kk] = v;
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
  const int64_t HW = (int64_t)cd.H * cd.W;
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int64_t m = bm + wm + i * 16 + crow0 + r;
        int ci = bn + wn + j * 16 + ccol;
        if (m < M && ci < cd.Cin) {
          const int iw = (int)(m % Wc);
          int64_t t = m / Wc;
          const int ih = (int)(t % Hb);
          const int n = (int)(t / Hb);
          dx[((int64_t)n * cd.Cin + ci) * HW +
             (int64_t)(2 * ih + b) * cd.W + (2 * iw + c)] =
              (__bf16)(acc[i][j][r]);
        }
      }
}

// ---------------------------------------------------------------------------
// Spatial-slab tap-reuse kernels (the conv3d_spatial.hip design with the
// depth axis dropped): a block stages one input slab [CTILE][H2][W2] per
// channel tile and computes the whole 9-tap K loop from it through a
// precomputed u16 k->offset table. Unlike the 3D kernels, staging is
// CLAMPED per row (vector loads only when the full interior is in
// bounds), so OWT need not divide the output width — ResNet's 28/14/7
// grids route here with partial edge tiles instead of falling back.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(8))) unsigned short u16x8_2;

struct Sp2Dims {
  int N, KCH;        // input-side channels (fwd: Cin; dgrad: Cout)
  int H, W;          // input-side spatial
  int NCOL;          // output-side channels
  int TH, TW;        // output-side spatial
  int Kpad;          // row length of WB = kts * KT_PAD
};

// NCOLT: output-channel tile per block. The deep small-image layers
// (7^2/14^2, 256-512 channels) are slab-staging bound — every 32-column
// block re-stages the same input slab; NCOLT=64 halves that redundancy
// and doubles the MFMA work per barrier.
template <int OWT, int STRIDE, int CTILE, int CHUNK, bool FUSE_BN = false,
          int NCOLT = 32>
__global__ __launch_bounds__(256) void conv2d_spatial_kernel(
    const __bf16* __restrict__ in, const __bf16* __restrict__ wb,
    __bf16* __restrict__ out, Sp2Dims sd, int64_t nchunks,
    const float* __restrict__ bn_ab) {
  constexpr int OHT = CHUNK / OWT;
  constexpr int IW = STRIDE * OWT;
  constexpr int W2 = IW + (STRIDE == 1 ? 4 : 2);
  constexpr int H2 = STRIDE * (OHT - 1) + 3;
  constexpr int MPW = CHUNK / 64;
  constexpr int NCF = NCOLT / 16;  // ncol fragments per wave
  static_assert(MPW >= 1, "chunk too small");
  constexpr int KT_PAD = ((CTILE * 9 + 31) / 32) * 32;
  __shared__ __bf16 sX[CTILE][H2][W2];
  __shared__ unsigned short sKtab[KT_PAD + 8];

  const int ncol0 = blockIdx.y * NCOLT;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int row = lane & 15, kg = lane >> 4;

  const int wtiles = (sd.TW + OWT - 1) / OWT;
  const int htiles = (sd.TH + OHT - 1) / OHT;

  int64_t t = blockIdx.x;
  const int wt = (int)(t % wtiles);
  t /= wtiles;
  const int ht = (int)(t % htiles);
  const int n = (int)(t / htiles);
  const int oh0 = ht * OHT, ow0 = wt * OWT;

  for (int k = tid; k < KT_PAD; k += 256) {
    unsigned short off = 0;
    if (k < CTILE * 9) {
      const int cl = k / 9;
      const int r9 = k - cl * 9;
      const int b = r9 / 3, c = r9 % 3;
      off = (unsigned short)((cl * H2 + b) * W2 + c);
    }
    sKtab[k] = off;
  }

  f32x4 acc[MPW][NCF];
#pragma unroll
  for (int i = 0; i < MPW; ++i)
#pragma unroll
    for (int j = 0; j < NCF; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)sd.H * sd.W;
  const int64_t in_n = (int64_t)n * sd.KCH * HW;
  const int kts = (sd.KCH + CTILE - 1) / CTILE;

  for (int kt = 0; kt < kts; ++kt) {
    constexpr int NROWS = CTILE * H2;
    if (kt) __syncthreads();
    for (int r = tid; r < NROWS; r += 256) {
      const int hrow = r % H2;
      const int c = r / H2;
      const int ih = STRIDE * oh0 - 1 + hrow;
      const int ch = kt * CTILE + c;
      __bf16* dst = &sX[c][hrow][0];
      const bool row_ok = (unsigned)ih < (unsigned)sd.H && ch < sd.KCH;
      if (!row_ok) {
#pragma unroll
        for (int col = 0; col < W2; ++col) dst[col] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = in + in_n + (int64_t)ch * HW + (int64_t)ih * sd.W;
      float a_c = 1.f, b_c = 0.f;
      if (FUSE_BN) { a_c = bn_ab[ch * 2]; b_c = bn_ab[ch * 2 + 1]; }
      auto tx = [&](__bf16 v) -> __bf16 {
        if (!FUSE_BN) return v;
        return (__bf16)fmaxf(a_c * (float)v + b_c, 0.f);
      };
      const int iw0 = STRIDE * ow0;
      if (iw0 + IW <= sd.W) {  // interior tile: vector loads
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
          dst[1 + IW + e] = (iw < sd.W) ? tx(src[iw]) : (__bf16)0.f;
        }
      } else {  // edge tile: clamped scalar loads
#pragma unroll
        for (int col = 0; col < W2; ++col) {
          const int iw = iw0 - 1 + col;
          dst[col] = ((unsigned)iw < (unsigned)sd.W) ? tx(src[iw])
                                                     : (__bf16)0.f;
        }
      }
    }
    __syncthreads();

    const int kbase_g = kt * KT_PAD;
#pragma unroll 1
    for (int ks = 0; ks < KT_PAD / 32; ++ks) {
      bf16x8 afrag[MPW];
      {
        const int kb = ks * 32 + kg * 8;
        const u16x8_2 kt8 = *reinterpret_cast<const u16x8_2*>(&sKtab[kb]);
        const __bf16* slab = &sX[0][0][0];
#pragma unroll
        for (int i = 0; i < MPW; ++i) {
          const int m = (wave * MPW + i) * 16 + row;
          const int base = (STRIDE * (m / OWT)) * W2 + STRIDE * (m % OWT);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            afrag[i][j] = slab[base + kt8[j]];
        }
      }
      bf16x8 bfrag[NCF];
#pragma unroll
      for (int i = 0; i < NCF; ++i) {
        const int col = ncol0 + i * 16 + row;
        const int64_t off =
            (int64_t)col * sd.Kpad + kbase_g + ks * 32 + kg * 8;
        bfrag[i] = (col < sd.NCOL)
                       ? *reinterpret_cast<const bf16x8*>(wb + off)
                       : bf16x8{};
      }
#pragma unroll
      for (int i = 0; i < MPW; ++i)
#pragma unroll
        for (int j = 0; j < NCF; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
  }

  const int64_t THW = (int64_t)sd.TH * sd.TW;
  const int64_t out_n = (int64_t)n * sd.NCOL * THW;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < MPW; ++i) {
#pragma unroll
    for (int j = 0; j < NCF; ++j) {
      const int col = ncol0 + j * 16 + ccol;
      if (col >= sd.NCOL) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = (wave * MPW + i) * 16 + crow0 + r;
        const int oh = oh0 + m / OWT;
        const int ow = ow0 + m % OWT;
        if (oh < sd.TH && ow < sd.TW)
          out[out_n + (int64_t)col * THW + (int64_t)oh * sd.TW + ow] =
              (__bf16)(acc[i][j][r]);
      }
    }
  }
}

// WGRAD with tap reuse (2D): block stages one x slab [CT][H2][W2] + one go
// tile [co][m] once and computes all 9 tap GEMMs from it; grid-strides
// over (n, h-tile, w-tile) chunks; fp32 atomics fold partials into dw.
// WIDE=true (stride 1 only): waves span 64 output channels (4x16) and
// each wave computes BOTH 16-wide ci fragments — halves the go-staging
// redundancy at the 256/512-channel stages and doubles the MFMA work
// per barrier (acc[9][2] = 72 VGPRs).
template <int OWT, int STRIDE, int CHUNK = 128, bool FUSE_BN = false,
          bool WIDE = false, bool SLICED = false>
__global__ __launch_bounds__(256) void conv2d_wgrad_sp_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ go,
    float* __restrict__ dw, Conv2dDims cd, int64_t nchunks, int64_t zstride,
    const float* __restrict__ bn_ab) {
  constexpr int OHT = CHUNK / OWT;
  constexpr int IW = STRIDE * OWT;
  constexpr int W2 = IW + (STRIDE == 1 ? 4 : 2);
  constexpr int H2 = STRIDE * (OHT - 1) + 3;
  constexpr int CT = STRIDE == 1 ? 32 : 16;
  constexpr int COT = (STRIDE == 1 && !WIDE) ? 32 : 64;
  constexpr int CIF = (STRIDE == 1 && WIDE) ? 2 : 1;  // ci frags per wave
  __shared__ __bf16 sX[CT][H2][W2];
  __shared__ __bf16 sGo[COT][CHUNK + LDA_PAD2];

  const int co0 = blockIdx.x * COT;
  const int ci0 = blockIdx.y * CT;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wi = (STRIDE == 1 && !WIDE) ? (wave >> 1) : wave;
  const int wj = (STRIDE == 1 && !WIDE) ? (wave & 1) : 0;
  const int row = lane & 15, kg = lane >> 4;

  const int wtiles = (cd.OW + OWT - 1) / OWT;
  const int htiles = (cd.OH + OHT - 1) / OHT;

  f32x4 acc[9][CIF];
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int f = 0; f < CIF; ++f) acc[t][f] = {0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)cd.H * cd.W;
  const int64_t OHW = (int64_t)cd.OH * cd.OW;

  for (int64_t z = blockIdx.z; z < nchunks; z += zstride) {
    int64_t t = z;
    const int wt = (int)(t % wtiles);
    t /= wtiles;
    const int ht = (int)(t % htiles);
    const int n = (int)(t / htiles);
    const int oh0 = ht * OHT, ow0 = wt * OWT;

    constexpr int NXROWS = CT * H2;
    const __bf16* xn = x + (int64_t)n * cd.Cin * HW;
    for (int r = tid; r < NXROWS; r += 256) {
      const int hrow = r % H2;
      const int ci = r / H2;
      const int ih = STRIDE * oh0 - 1 + hrow;
      __bf16* dst = &sX[ci][hrow][0];
      const bool row_ok = (unsigned)ih < (unsigned)cd.H &&
                          (ci0 + ci) < cd.Cin;
      if (!row_ok) {
#pragma unroll
        for (int col = 0; col < W2; ++col) dst[col] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = xn + (int64_t)(ci0 + ci) * HW +
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
      if (iw0 + IW <= cd.W) {
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
      } else {
#pragma unroll
        for (int col = 0; col < W2; ++col) {
          const int iw = iw0 - 1 + col;
          dst[col] = ((unsigned)iw < (unsigned)cd.W) ? tx(src[iw])
                                                     : (__bf16)0.f;
        }
      }
    }
    const __bf16* gon = go + (int64_t)n * cd.Cout * OHW;
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
      const __bf16* src = gon + (int64_t)(co0 + co) * OHW +
                          (int64_t)oh * cd.OW + ow0;
      if (ow0 + OWT <= cd.OW) {
#pragma unroll
        for (int v = 0; v < OWT / 8; ++v)
          *reinterpret_cast<bf16x8*>(dst + v * 8) =
              *reinterpret_cast<const bf16x8*>(src + v * 8);
      } else {
#pragma unroll
        for (int j = 0; j < OWT; ++j)
          dst[j] = (ow0 + j < cd.OW) ? src[j] : (__bf16)0.f;
      }
    }
    __syncthreads();

#pragma unroll 1
    for (int ms = 0; ms < CHUNK / 32; ++ms) {
      bf16x8 afrag;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        afrag[j] = sGo[wi * 16 + row][ms * 32 + kg * 8 + j];
      const int mbase = ms * 32 + kg * 8;
      const int oh_off = mbase / OWT;
      const int ow_off = mbase % OWT;
#pragma unroll
      for (int kh = 0; kh < 3; ++kh) {
#pragma unroll
        for (int kw = 0; kw < 3; ++kw) {
#pragma unroll
          for (int f = 0; f < CIF; ++f) {
            bf16x8 bfrag;
            const __bf16* src = &sX[(wj + f) * 16 + row]
                                   [STRIDE * oh_off + kh]
                                   [STRIDE * ow_off + kw];
#pragma unroll
            for (int j = 0; j < 8; ++j) bfrag[j] = src[STRIDE * j];
            acc[kh * 3 + kw][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag, bfrag, acc[kh * 3 + kw][f], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  const int K = cd.Cin * 9;
  float* out = SLICED ? dw + (int64_t)blockIdx.z * cd.Cout * K : dw;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll 1
  for (int tp = 0; tp < 9; ++tp) {
#pragma unroll
    for (int f = 0; f < CIF; ++f) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = co0 + wi * 16 + crow0 + r;
        const int ci = ci0 + (wj + f) * 16 + ccol;
        if (co < cd.Cout && ci < cd.Cin) {
          if (SLICED)
            out[(int64_t)co * K + ci * 9 + tp] = acc[tp][f][r];
          else
            atomicAdd(&out[(int64_t)co * K + ci * 9 + tp], acc[tp][f][r]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// WGRAD: split-K implicit GEMM with fp32 atomics (structure of
// conv3d.hip's conv3d_wgrad_kernel with the depth axis dropped).
// ---------------------------------------------------------------------------
#define WMB2 128

template <bool FUSE_BN = false, int KS = 3>
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

  constexpr int KK = KS * KS;
  constexpr int PAD = KS / 2;
  const int mi0 = (tid & 7) * 4;
  const int ct = tid >> 3;
  const int kq = k0 + ct;
  const int ci = kq / KK;
  const int rr = kq - ci * KK;
  const int kh = rr / KS, kw = rr % KS;
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
          const int ih = jh * cd.stride - PAD + kh;
          const int iw = jw * cd.stride - PAD + kw;
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
                             int stride, int ks = 3) {
  Conv2dDims cd;
  cd.N = (int)x.size(0); cd.Cin = (int)x.size(1);
  cd.H = (int)x.size(2); cd.W = (int)x.size(3);
  cd.Cout = (int)w.size(0);
  cd.stride = stride;
  cd.OH = (cd.H + 2 * (ks / 2) - ks) / stride + 1;
  cd.OW = (cd.W + 2 * (ks / 2) - ks) / stride + 1;
  return cd;
}

// WB layout: [NCOL][kts * KT_PAD]; channel-tile block kt holds the CTILE*9
// weights, zero-padded to a 32-multiple (CTILE=32 -> 288, no padding).
static torch::Tensor prep_wb2(torch::Tensor w_flat2d, int KCH, int ctile) {
  int ncol = (int)w_flat2d.size(0);
  int kt_pad = (ctile * 9 + 31) / 32 * 32;
  int kts = (KCH + ctile - 1) / ctile;
  auto wb = torch::zeros({ncol, (int64_t)kts * kt_pad}, w_flat2d.options());
  for (int kt = 0; kt < kts; ++kt) {
    int64_t k0 = (int64_t)kt * ctile * 9;
    int64_t klen = (int64_t)std::min(KCH - kt * ctile, ctile) * 9;
    wb.narrow(1, (int64_t)kt * kt_pad, klen).copy_(
        w_flat2d.narrow(1, k0, klen));
  }
  return wb;
}

static int pick_owt2(int tw) {
  if (tw % 32 == 0) return 32;
  if (tw % 16 == 0) return 16;
  return 8;  // clamped staging: divisibility not required
}

static void launch_spatial2(torch::Tensor in, torch::Tensor wb,
                            torch::Tensor out, Sp2Dims sd, int stride,
                            const float* bn_ab) {
  int OWT = pick_owt2(sd.TW);
  int chunk = stride == 1 ? 256 : 128;
  int ncolt = 32;
  if (sd.TH * sd.TW < chunk) {
    chunk = 64;
    OWT = 8;
    // wide columns cut slab re-reads; 128 won the 3D sweep at deep layers
    if (sd.NCOL >= 64) ncolt = sd.NCOL >= 128 ? 128 : 64;
  } else if (stride == 1 && sd.NCOL >= 64 && bn_ab == nullptr) {
    ncolt = 64;  // non-fused chunk-256 wide (same win as 3D: dgrad path)
  }
  int OHT = chunk / OWT;
  int wtiles = (sd.TW + OWT - 1) / OWT;
  int htiles = (sd.TH + OHT - 1) / OHT;
  int64_t nchunks = (int64_t)sd.N * htiles * wtiles;
  dim3 grid((unsigned)nchunks, (sd.NCOL + ncolt - 1) / ncolt);
  auto s = current_stream();
  const __bf16* ip = reinterpret_cast<const __bf16*>(in.data_ptr());
  const __bf16* wp = reinterpret_cast<const __bf16*>(wb.data_ptr());
  __bf16* op = reinterpret_cast<__bf16*>(out.data_ptr());
  auto L = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, s, ip, wp, op, sd, nchunks,
                       bn_ab);
  };
  if (bn_ab != nullptr) {
    if (stride == 1) {
      if (chunk == 64 && ncolt == 128)
        L(conv2d_spatial_kernel<8, 1, 32, 64, true, 128>);
      else if (chunk == 64 && ncolt == 64)
        L(conv2d_spatial_kernel<8, 1, 32, 64, true, 64>);
      else if (chunk == 64) L(conv2d_spatial_kernel<8, 1, 32, 64, true>);
      else if (OWT == 32) L(conv2d_spatial_kernel<32, 1, 32, 256, true>);
      else if (OWT == 16) L(conv2d_spatial_kernel<16, 1, 32, 256, true>);
      else L(conv2d_spatial_kernel<8, 1, 32, 256, true>);
    } else {
      if (chunk == 64 && ncolt == 128)
        L(conv2d_spatial_kernel<8, 2, 16, 64, true, 128>);
      else if (chunk == 64 && ncolt == 64)
        L(conv2d_spatial_kernel<8, 2, 16, 64, true, 64>);
      else if (chunk == 64) L(conv2d_spatial_kernel<8, 2, 16, 64, true>);
      else if (OWT == 32) L(conv2d_spatial_kernel<32, 2, 16, 128, true>);
      else if (OWT == 16) L(conv2d_spatial_kernel<16, 2, 16, 128, true>);
      else L(conv2d_spatial_kernel<8, 2, 16, 128, true>);
    }
  } else if (stride == 1) {
    if (chunk == 64 && ncolt == 128)
      L(conv2d_spatial_kernel<8, 1, 32, 64, false, 128>);
    else if (chunk == 64 && ncolt == 64)
      L(conv2d_spatial_kernel<8, 1, 32, 64, false, 64>);
    else if (chunk == 64) L(conv2d_spatial_kernel<8, 1, 32, 64>);
    else if (OWT == 32 && ncolt == 64)
      L(conv2d_spatial_kernel<32, 1, 32, 256, false, 64>);
    else if (OWT == 32) L(conv2d_spatial_kernel<32, 1, 32, 256>);
    else if (OWT == 16 && ncolt == 64)
      L(conv2d_spatial_kernel<16, 1, 32, 256, false, 64>);
    else if (OWT == 16) L(conv2d_spatial_kernel<16, 1, 32, 256>);
    else if (ncolt == 64)
      L(conv2d_spatial_kernel<8, 1, 32, 256, false, 64>);
    else L(conv2d_spatial_kernel<8, 1, 32, 256>);
  } else {
    if (chunk == 64 && ncolt == 128)
      L(conv2d_spatial_kernel<8, 2, 16, 64, false, 128>);
    else if (chunk == 64 && ncolt == 64)
      L(conv2d_spatial_kernel<8, 2, 16, 64, false, 64>);
    else if (chunk == 64) L(conv2d_spatial_kernel<8, 2, 16, 64>);
    else if (OWT == 32) L(conv2d_spatial_kernel<32, 2, 16, 128>);
    else if (OWT == 16) L(conv2d_spatial_kernel<16, 2, 16, 128>);
    else L(conv2d_spatial_kernel<8, 2, 16, 128>);
  }
}

torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, int64_t stride,
                         c10::optional<torch::Tensor> bn_ab_opt) {
  torch::Tensor bn_ab = bn_ab_opt.value_or(torch::Tensor());
  CHECK_GPU(x);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "conv2d_fwd wants bf16");
  auto xc = x.contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  const int ks = (int)wc.size(2);
  TORCH_CHECK(wc.size(3) == ks && (ks == 3 || ks == 7),
              "3x3 or 7x7 kernels only");
  auto cd = make_dims2(xc, wc, (int)stride, ks);
  TORCH_CHECK(wc.size(1) == cd.Cin, "channel mismatch");
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

  if (ks == 3 && cd.Cin >= 16 && cd.OH * cd.OW >= 32) {
    // spatial-slab tap-reuse path (clamped staging: any width)
    Sp2Dims sd;
    sd.N = cd.N; sd.KCH = cd.Cin; sd.H = cd.H; sd.W = cd.W;
    sd.NCOL = cd.Cout; sd.TH = cd.OH; sd.TW = cd.OW;
    int ctile = stride == 1 ? 32 : 16;
    auto wb = prep_wb2(wc.reshape({sd.NCOL, (int64_t)sd.KCH * 9}), sd.KCH,
                       ctile);
    sd.Kpad = (int)wb.size(1);
    launch_spatial2(xc, wb, out, sd, (int)stride, abp);
    return out;
  }

  int64_t M = (int64_t)cd.N * cd.OH * cd.OW;
  int K = cd.Cin * ks * ks;
  dim3 grid((unsigned)((M + CBM2 - 1) / CBM2), (cd.Cout + CBN2 - 1) / CBN2);
  auto L = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<const __bf16*>(wc.data_ptr()),
                       reinterpret_cast<__bf16*>(out.data_ptr()), cd, M,
                       cd.Cout, K, abp);
  };
  if (ks == 7) {
    TORCH_CHECK(!fuse, "7x7 fwd: no fused-BN instances");
    if (stride == 1) L(conv2d_igemm_kernel<false, 1, false, 7>);
    else L(conv2d_igemm_kernel<false, 2, false, 7>);
  } else if (fuse) {
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

  if (stride == 1 && cd.Cout >= 16 && cd.H * cd.W >= 32) {
    // dgrad as spatial conv of go with tap-flipped transposed weights
    Sp2Dims sd;
    sd.N = cd.N; sd.KCH = cd.Cout; sd.H = cd.OH; sd.W = cd.OW;
    sd.NCOL = cd.Cin; sd.TH = cd.H; sd.TW = cd.W;
    auto wf = wc.reshape({cd.Cout, cd.Cin, 9}).flip(-1).permute({1, 0, 2})
                  .reshape({cd.Cin, (int64_t)cd.Cout * 9}).contiguous();
    auto wb = prep_wb2(wf, cd.Cout, 32);
    sd.Kpad = (int)wb.size(1);
    launch_spatial2(g, wb, dx, sd, 1, nullptr);
    return dx;
  }

  if (stride == 2) {
    // parity classes: dense sub-GEMMs, M = worst-case class size
    int64_t Mmax = (int64_t)cd.N * ((cd.H + 1) / 2) * ((cd.W + 1) / 2);
    dim3 grid((unsigned)((Mmax + CBM2 - 1) / CBM2),
              (cd.Cin + CBN2 - 1) / CBN2, 4);
    hipLaunchKernelGGL(conv2d_dgrad_s2_kernel, grid, dim3(256), 0,
                       current_stream(),
                       reinterpret_cast<const __bf16*>(g.data_ptr()),
                       reinterpret_cast<const __bf16*>(wc.data_ptr()),
                       reinterpret_cast<__bf16*>(dx.data_ptr()), cd);
    return dx;
  }
  int64_t M = (int64_t)cd.N * cd.H * cd.W;
  int K = cd.Cout * 9;
  dim3 grid((unsigned)((M + CBM2 - 1) / CBM2), (cd.Cin + CBN2 - 1) / CBN2);
  hipLaunchKernelGGL((conv2d_igemm_kernel<true, 1>), grid, dim3(256), 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(g.data_ptr()),
                     reinterpret_cast<const __bf16*>(wc.data_ptr()),
                     reinterpret_cast<__bf16*>(dx.data_ptr()), cd, M,
                     cd.Cin, K, (const float*)nullptr);
  return dx;
}

torch::Tensor conv2d_wgrad(torch::Tensor x, torch::Tensor go,
                           int64_t stride, c10::optional<torch::Tensor> bn_ab_opt,
                           int64_t ks) {
  torch::Tensor bn_ab = bn_ab_opt.value_or(torch::Tensor());
  TORCH_CHECK(ks == 3 || ks == 7, "3x3 or 7x7 kernels only");
  CHECK_GPU(x);
  auto xc = x.to(torch::kBFloat16).contiguous();
  auto g = go.to(torch::kBFloat16).contiguous();
  Conv2dDims cd;
  cd.N = (int)xc.size(0); cd.Cin = (int)xc.size(1);
  cd.H = (int)xc.size(2); cd.W = (int)xc.size(3);
  cd.stride = (int)stride;
  cd.Cout = (int)g.size(1);
  cd.OH = (int)g.size(2); cd.OW = (int)g.size(3);
  int K = cd.Cin * (int)(ks * ks);
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

  if (ks == 3 && cd.Cin >= 16 && cd.OH * cd.OW >= 32 &&
      (stride == 1 || stride == 2)) {
    int OWT = pick_owt2(cd.OW);
    int chunk = 128;
    if (cd.OH * cd.OW < 128) { chunk = 64; OWT = 8; }
    int wtiles = (cd.OW + OWT - 1) / OWT;
    int OHT = chunk / OWT;
    int htiles = (cd.OH + OHT - 1) / OHT;
    // wide measured WORSE on ResNet (18.3 vs 15.7 ms/step, r2 A/B):
    // the halved z-parallelism + doubled sGo cost more than the halved
    // staging redundancy buys. Instances stay compiled; routing off.
    const bool wide = false && (stride == 1 && cd.Cout >= 64 && !fuse);
    int COT = stride == 1 ? (wide ? 64 : 32) : 64;
    int CT = stride == 1 ? 32 : 16;
    int co_t = (cd.Cout + COT - 1) / COT, ci_t = (cd.Cin + CT - 1) / CT;
    int64_t nchunks = (int64_t)cd.N * htiles * wtiles;
    int64_t zstride = std::max<int64_t>(
        1, std::min<int64_t>(nchunks, 768 / std::max(co_t * ci_t, 1)));
    dim3 grid(co_t, ci_t, (unsigned)zstride);
    const bool sliced = zstride >= 16 &&
                        (int64_t)zstride * cd.Cout * K * 4 <=
                            (int64_t)512 * 1024 * 1024;
    torch::Tensor part;
    float* outp = dw.data_ptr<float>();
    if (sliced) {
      part = torch::empty({(int64_t)zstride, (int64_t)cd.Cout, (int64_t)K},
                          xc.options().dtype(torch::kFloat32));
      outp = part.data_ptr<float>();
    }
    auto L = [&](auto kern) {
      hipLaunchKernelGGL(kern, grid, dim3(256), 0, current_stream(),
                         reinterpret_cast<const __bf16*>(xc.data_ptr()),
                         reinterpret_cast<const __bf16*>(g.data_ptr()),
                         outp, cd, nchunks, zstride, abp);
    };
    if (sliced) {
      if (fuse) {
        if (stride == 1) {
          if (chunk == 64)
            L(conv2d_wgrad_sp_kernel<8, 1, 64, true, false, true>);
          else if (OWT == 32)
            L(conv2d_wgrad_sp_kernel<32, 1, 128, true, false, true>);
          else if (OWT == 16)
            L(conv2d_wgrad_sp_kernel<16, 1, 128, true, false, true>);
          else L(conv2d_wgrad_sp_kernel<8, 1, 128, true, false, true>);
        } else {
          if (chunk == 64)
            L(conv2d_wgrad_sp_kernel<8, 2, 64, true, false, true>);
          else if (OWT == 32)
            L(conv2d_wgrad_sp_kernel<32, 2, 128, true, false, true>);
          else if (OWT == 16)
            L(conv2d_wgrad_sp_kernel<16, 2, 128, true, false, true>);
          else L(conv2d_wgrad_sp_kernel<8, 2, 128, true, false, true>);
        }
      } else if (stride == 1) {
        if (chunk == 64)
          L(conv2d_wgrad_sp_kernel<8, 1, 64, false, false, true>);
        else if (OWT == 32)
          L(conv2d_wgrad_sp_kernel<32, 1, 128, false, false, true>);
        else if (OWT == 16)
          L(conv2d_wgrad_sp_kernel<16, 1, 128, false, false, true>);
        else L(conv2d_wgrad_sp_kernel<8, 1, 128, false, false, true>);
      } else {
        if (chunk == 64)
          L(conv2d_wgrad_sp_kernel<8, 2, 64, false, false, true>);
        else if (OWT == 32)
          L(conv2d_wgrad_sp_kernel<32, 2, 128, false, false, true>);
        else if (OWT == 16)
          L(conv2d_wgrad_sp_kernel<16, 2, 128, false, false, true>);
        else L(conv2d_wgrad_sp_kernel<8, 2, 128, false, false, true>);
      }
      return part.sum(0).view({cd.Cout, cd.Cin, 3, 3});
    }
    if (fuse) {
      // fused-BN wgrad keeps the 32-wide form (no wide instances)
      if (stride == 1) {
        if (chunk == 64) L(conv2d_wgrad_sp_kernel<8, 1, 64, true>);
        else if (OWT == 32) L(conv2d_wgrad_sp_kernel<32, 1, 128, true>);
        else if (OWT == 16) L(conv2d_wgrad_sp_kernel<16, 1, 128, true>);
        else L(conv2d_wgrad_sp_kernel<8, 1, 128, true>);
      } else {
        if (chunk == 64) L(conv2d_wgrad_sp_kernel<8, 2, 64, true>);
        else if (OWT == 32) L(conv2d_wgrad_sp_kernel<32, 2, 128, true>);
        else if (OWT == 16) L(conv2d_wgrad_sp_kernel<16, 2, 128, true>);
        else L(conv2d_wgrad_sp_kernel<8, 2, 128, true>);
      }
    } else if (stride == 1 && wide) {
      if (chunk == 64) L(conv2d_wgrad_sp_kernel<8, 1, 64, false, true>);
      else if (OWT == 32) L(conv2d_wgrad_sp_kernel<32, 1, 128, false, true>);
      else if (OWT == 16) L(conv2d_wgrad_sp_kernel<16, 1, 128, false, true>);
      else L(conv2d_wgrad_sp_kernel<8, 1, 128, false, true>);
    } else if (stride == 1) {
      if (chunk == 64) L(conv2d_wgrad_sp_kernel<8, 1, 64>);
      else if (OWT == 32) L(conv2d_wgrad_sp_kernel<32, 1, 128>);
      else if (OWT == 16) L(conv2d_wgrad_sp_kernel<16, 1, 128>);
      else L(conv2d_wgrad_sp_kernel<8, 1, 128>);
    } else {
      if (chunk == 64) L(conv2d_wgrad_sp_kernel<8, 2, 64>);
      else if (OWT == 32) L(conv2d_wgrad_sp_kernel<32, 2, 128>);
      else if (OWT == 16) L(conv2d_wgrad_sp_kernel<16, 2, 128>);
      else L(conv2d_wgrad_sp_kernel<8, 2, 128>);
    }
    return dw.view({cd.Cout, cd.Cin, 3, 3});
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
  if (ks == 7) {
    TORCH_CHECK(!fuse, "7x7 wgrad: no fused-BN instances");
    L(conv2d_wgrad_kernel<false, 7>);
  } else if (fuse) {
    L(conv2d_wgrad_kernel<true>);
  } else {
    L(conv2d_wgrad_kernel<false>);
  }
  return dw.view({cd.Cout, cd.Cin, (int64_t)ks, (int64_t)ks});
}
