// Conv3d fwd/dgrad with spatial-slab tap reuse (stride 1 and stride-2 fwd).
// The igemm formulation re-gathers x once per (ci,tap) — 27x traffic.
// Here a block stages one input spatial slab [CTILE][3][H2][W2] per
// channel tile and computes the whole K loop from it:
//   - slab rows are staged with 16-byte vector loads (interior columns are
//     in-bounds by construction since OWT divides the output width);
//   - the A fragments read the slab at base(m) + sKtab[k], a precomputed
//     u16 offset table (tap decode off the hot path);
//   - the B fragments are contiguous 16-byte global loads from the small
//     L2-resident prepared-weight matrix WB[ncol][kts*KT_PAD] (fwd: w
//     reshaped; dgrad: tap-flipped + transposed; each CTILE-channel block
//     zero-padded to a 32-multiple so MFMA k-windows never straddle tiles).
// Geometry: out tile = (OWT*OHT) m x 32 ncol; 4 waves x MPW m-fragments x
// 2 ncol-fragments; chunk = (n, d, h-tile, w-tile) of the OUTPUT grid.
// STRIDE=2 stages the strided window (IW = 2*OWT interior) with CTILE=16.
#include "common.h"

#include <hip/hip_bf16.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;

struct SpDims {
  int N, KCH;        // input-side channels (fwd: Cin; dgrad: Cout)
  int D, H, W;       // input-side spatial (fwd: x dims; dgrad: go dims)
  int NCOL;          // output-side channels
  int TD, TH, TW;    // output-side spatial
  int Kpad;          // row length of WB = kts * KT_PAD
};

#ifndef LDA_PAD
#define LDA_PAD 8
#endif

// FUSE_BN: normalize-on-load — the staged input is the PREVIOUS block's
// raw conv output; z = relu(a[c]*x + b[c]) is applied per element during
// staging (a = gamma*rstd, b = beta - mean*a, exactly bn_normalize's
// folding) so the normalized tensor never exists in HBM. Padding halo
// stays 0 (conv pads the BN OUTPUT with zeros).
// SMODE 1: the epilogue also folds per-output-channel (sum, sumsq) of
// the written elements into stats[hash_slice][NCOL][2] — the NEXT
// block's BN statistics come for free (no standalone bn_reduce pass).
// SMODE 2 (dgrad use): the written elements are dz of a BN(+ReLU)
// output; fold the BN-BACKWARD reduction (sum(dz*mask), sum(dz*xhat*
// mask)) instead, reading x_raw at the same offsets with the mask
// recomputed from bn_prm = [mean, rstd, gamma, beta] per channel —
// replaces the standalone bn_bwd_reduce pass.
// Hash slices (blockIdx.x & 63) spread the atomics.
template <int OWT, int STRIDE, int CTILE,
          int CHUNK = (STRIDE == 1 ? 256 : 128), bool FUSE_BN = false,
          int SMODE = 0, int NCOLT = 32>
__global__ __launch_bounds__(256) void conv3d_spatial_kernel(
    const __bf16* __restrict__ in, const __bf16* __restrict__ wb,
    __bf16* __restrict__ out, SpDims sd, int64_t nchunks,
    const float* __restrict__ bn_ab = nullptr,
    float* __restrict__ stats = nullptr,
    const __bf16* __restrict__ xraw = nullptr,
    const float* __restrict__ bn_prm = nullptr) {
  constexpr int OHT = CHUNK / OWT;
  constexpr int IW = STRIDE * OWT;                  // staged interior width
  constexpr int W2 = IW + (STRIDE == 1 ? 4 : 2);
  constexpr int H2 = STRIDE * (OHT - 1) + 3;
  constexpr int MPW = CHUNK / 64;                   // m-frags per wave
  static_assert(MPW >= 1, "chunk too small");
  constexpr int KT_PAD = ((CTILE * 27 + 31) / 32) * 32;
  constexpr int NCF = NCOLT / 16;  // ncol fragments per wave
  __shared__ __bf16 sX[CTILE][3][H2][W2];
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
  t /= htiles;
  const int td = (int)(t % sd.TD);
  const int n = (int)(t / sd.TD);
  const int oh0 = ht * OHT, ow0 = wt * OWT;

  // k -> slab element offset (pad entries -> 0: their weights are zero)
  for (int k = tid; k < KT_PAD; k += 256) {
    unsigned short off = 0;
    if (k < CTILE * 27) {
      const int cl = k / 27;
      const int r27 = k - cl * 27;
      const int a = r27 / 9, b = (r27 / 3) % 3, c = r27 % 3;
      off = (unsigned short)(((cl * 3 + a) * H2 + b) * W2 + c);
    }
    sKtab[k] = off;
  }

  f32x4 acc[MPW][NCF];
#pragma unroll
  for (int i = 0; i < MPW; ++i)
#pragma unroll
    for (int j = 0; j < NCF; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)sd.H * sd.W;
  const int64_t in_n = (int64_t)n * sd.KCH * sd.D * HW;
  const int kts = (sd.KCH + CTILE - 1) / CTILE;

  for (int kt = 0; kt < kts; ++kt) {
    // ---- stage the slab row-wise (16B interior, scalar halo) ----------
    constexpr int NROWS = CTILE * 3 * H2;
    if (kt) __syncthreads();
    for (int r = tid; r < NROWS; r += 256) {
      const int hrow = r % H2;
      const int a = (r / H2) % 3;
      const int c = r / (3 * H2);
      const int id = STRIDE * td - 1 + a;
      const int ih = STRIDE * oh0 - 1 + hrow;
      const int ch = kt * CTILE + c;
      __bf16* dst = &sX[c][a][hrow][0];
      const bool row_ok = (unsigned)id < (unsigned)sd.D &&
                          (unsigned)ih < (unsigned)sd.H && ch < sd.KCH;
      if (!row_ok) {
#pragma unroll
        for (int col = 0; col < W2; ++col) dst[col] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = in + in_n + ((int64_t)ch * sd.D + id) * HW +
                          (int64_t)ih * sd.W;
      float a_c = 1.f, b_c = 0.f;
      if (FUSE_BN) { a_c = bn_ab[ch * 2]; b_c = bn_ab[ch * 2 + 1]; }
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
        dst[1 + IW + e] = (iw < sd.W) ? tx(src[iw]) : (__bf16)0.f;
      }
    }
    __syncthreads();

    // ---- k-steps of 32 over (ch_local, tap) ---------------------------
    const int kbase_g = kt * KT_PAD;
#pragma unroll 1
    for (int ks = 0; ks < KT_PAD / 32; ++ks) {
      bf16x8 afrag[MPW];
      {
        const int kb = ks * 32 + kg * 8;
        const u16x8 kt8 = *reinterpret_cast<const u16x8*>(&sKtab[kb]);
        const __bf16* slab = &sX[0][0][0][0];
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

  // ---- epilogue: out[n][col][td][oh0+][ow0+] --------------------------
  const int64_t THW = (int64_t)sd.TH * sd.TW;
  const int64_t out_n = (int64_t)n * sd.NCOL * sd.TD * THW;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  float cs[8] = {}, css[8] = {};  // per-col-frag partials (NCF <= 8)
  static_assert(NCF <= 8, "stats buffers sized for NCF <= 8");
#pragma unroll
  for (int i = 0; i < MPW; ++i) {
#pragma unroll
    for (int j = 0; j < NCF; ++j) {
      const int col = ncol0 + j * 16 + ccol;
      if (col >= sd.NCOL) continue;
      float p_mean = 0.f, p_rstd = 0.f, p_g = 0.f, p_b = 0.f;
      if (SMODE == 2) {
        p_mean = bn_prm[col * 4 + 0];
        p_rstd = bn_prm[col * 4 + 1];
        p_g = bn_prm[col * 4 + 2];
        p_b = bn_prm[col * 4 + 3];
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = (wave * MPW + i) * 16 + crow0 + r;
        const int oh = oh0 + m / OWT;
        const int ow = ow0 + m % OWT;
        if (oh < sd.TH && ow < sd.TW) {
          const float v = acc[i][j][r];
          const int64_t off = out_n + ((int64_t)col * sd.TD + td) * THW +
                              (int64_t)oh * sd.TW + ow;
          out[off] = (__bf16)v;
          if (SMODE == 1) {
            // accumulate the ROUNDED value: bit-matches bn3d_stats on
            // the stored bf16 tensor
            const float vb = (float)(__bf16)v;
            cs[j] += vb;
            css[j] += vb * vb;
          } else if (SMODE == 2) {
            const float xh = ((float)xraw[off] - p_mean) * p_rstd;
            if (p_g * xh + p_b > 0.f) {  // ReLU mask
              const float d = (float)(__bf16)v;
              cs[j] += d;
              css[j] += d * xh;
            }
          }
        }
      }
    }
  }
  if (SMODE != 0) {
    // lanes l, l+16, l+32, l+48 share ccol: fold over lane bits 4-5
#pragma unroll
    for (int j = 0; j < NCF; ++j) {
      cs[j] += __shfl_xor(cs[j], 16);
      cs[j] += __shfl_xor(cs[j], 32);
      css[j] += __shfl_xor(css[j], 16);
      css[j] += __shfl_xor(css[j], 32);
    }
    __shared__ float sred[4][NCF][16][2];
    if ((lane >> 4) == 0) {
#pragma unroll
      for (int j = 0; j < NCF; ++j) {
        sred[wave][j][ccol][0] = cs[j];
        sred[wave][j][ccol][1] = css[j];
      }
    }
    __syncthreads();
    // NCF*32 (col, stat) pairs: low threads fold the 4 waves and emit
    // one atomicAdd each into the hashed stats slice
    if (tid < NCF * 32) {
      const int j = tid >> 5;           // col fragment
      const int cc = (tid >> 1) & 15;   // ccol
      const int st = tid & 1;           // 0 = sum, 1 = sumsq
      const int col = ncol0 + j * 16 + cc;
      if (col < sd.NCOL) {
        float t = 0.f;
#pragma unroll
        for (int w = 0; w < 4; ++w) t += sred[w][j][cc][st];
        float* slice = stats +
            ((int64_t)(blockIdx.x & 63) * sd.NCOL + col) * 2 + st;
        atomicAdd(slice, t);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Double-buffered variant (experimental; routed only via ctile_opt=16):
// stage slab kt+1 into the spare LDS buffer while the 14 k-steps of slab
// kt run, one barrier per channel tile instead of two. CTILE=16 so two
// buffers fit LDS. Same math, same WB layout (KT_PAD blocks at ctile 16).
// ---------------------------------------------------------------------------
template <int OWT, int STRIDE, int CTILE,
          int CHUNK = (STRIDE == 1 ? 256 : 128)>
__global__ __launch_bounds__(256) void conv3d_spatial_db_kernel(
    const __bf16* __restrict__ in, const __bf16* __restrict__ wb,
    __bf16* __restrict__ out, SpDims sd, int64_t nchunks) {
  constexpr int NCF = 2;  // fixed 32-col tiles in the DB variant
  constexpr int OHT = CHUNK / OWT;
  constexpr int IW = STRIDE * OWT;
  constexpr int W2 = IW + (STRIDE == 1 ? 4 : 2);
  constexpr int H2 = STRIDE * (OHT - 1) + 3;
  constexpr int MPW = CHUNK / 64;
  static_assert(MPW >= 1, "chunk too small");
  constexpr int KT_PAD = ((CTILE * 27 + 31) / 32) * 32;
  __shared__ __bf16 sX[2][CTILE][3][H2][W2];
  __shared__ unsigned short sKtab[KT_PAD + 8];

  const int ncol0 = blockIdx.y * 32;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int row = lane & 15, kg = lane >> 4;

  const int wtiles = (sd.TW + OWT - 1) / OWT;
  const int htiles = (sd.TH + OHT - 1) / OHT;

  int64_t t = blockIdx.x;
  const int wt = (int)(t % wtiles);
  t /= wtiles;
  const int ht = (int)(t % htiles);
  t /= htiles;
  const int td = (int)(t % sd.TD);
  const int n = (int)(t / sd.TD);
  const int oh0 = ht * OHT, ow0 = wt * OWT;

  for (int k = tid; k < KT_PAD; k += 256) {
    unsigned short off = 0;
    if (k < CTILE * 27) {
      const int cl = k / 27;
      const int r27 = k - cl * 27;
      const int a = r27 / 9, b = (r27 / 3) % 3, c = r27 % 3;
      off = (unsigned short)(((cl * 3 + a) * H2 + b) * W2 + c);
    }
    sKtab[k] = off;
  }

  f32x4 acc[MPW][NCF];
#pragma unroll
  for (int i = 0; i < MPW; ++i)
#pragma unroll
    for (int j = 0; j < NCF; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)sd.H * sd.W;
  const int64_t in_n = (int64_t)n * sd.KCH * sd.D * HW;
  const int kts = (sd.KCH + CTILE - 1) / CTILE;

  auto stage = [&](int kt, int buf) {
    constexpr int NROWS = CTILE * 3 * H2;
    for (int r = tid; r < NROWS; r += 256) {
      const int hrow = r % H2;
      const int a = (r / H2) % 3;
      const int c = r / (3 * H2);
      const int id = STRIDE * td - 1 + a;
      const int ih = STRIDE * oh0 - 1 + hrow;
      const int ch = kt * CTILE + c;
      __bf16* dst = &sX[buf][c][a][hrow][0];
      const bool row_ok = (unsigned)id < (unsigned)sd.D &&
                          (unsigned)ih < (unsigned)sd.H && ch < sd.KCH;
      if (!row_ok) {
#pragma unroll
        for (int col = 0; col < W2; ++col) dst[col] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = in + in_n + ((int64_t)ch * sd.D + id) * HW +
                          (int64_t)ih * sd.W;
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
        dst[1 + IW + e] = (iw < sd.W) ? src[iw] : (__bf16)0.f;
      }
    }
  };

  auto compute = [&](int kt, int buf) {
    const int kbase_g = kt * KT_PAD;
    const __bf16* slab = &sX[buf][0][0][0][0];
#pragma unroll 1
    for (int ks = 0; ks < KT_PAD / 32; ++ks) {
      bf16x8 afrag[MPW];
      {
        const int kb = ks * 32 + kg * 8;
        const u16x8 kt8 = *reinterpret_cast<const u16x8*>(&sKtab[kb]);
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
  };

  stage(0, 0);
  __syncthreads();
  for (int kt = 0; kt < kts; ++kt) {
    if (kt + 1 < kts) stage(kt + 1, (kt + 1) & 1);
    compute(kt, kt & 1);
    __syncthreads();  // next buffer staged AND this buffer's reads done
  }

  const int64_t THW = (int64_t)sd.TH * sd.TW;
  const int64_t out_n = (int64_t)n * sd.NCOL * sd.TD * THW;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < MPW; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int col = ncol0 + j * 16 + ccol;
      if (col >= sd.NCOL) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = (wave * MPW + i) * 16 + crow0 + r;
        const int oh = oh0 + m / OWT;
        const int ow = ow0 + m % OWT;
        if (oh < sd.TH && ow < sd.TW)
          out[out_n + ((int64_t)col * sd.TD + td) * THW +
              (int64_t)oh * sd.TW + ow] = (__bf16)(acc[i][j][r]);
      }
    }
  }
}

// ---- host -----------------------------------------------------------------
// WB layout: [NCOL][kts * KT_PAD]; channel-tile block kt holds the CTILE*27
// weights for channels [kt*CTILE, (kt+1)*CTILE), zero-padded to KT_PAD.
static torch::Tensor prep_wb(torch::Tensor w_flat2d, int KCH, int ctile) {
  int ncol = (int)w_flat2d.size(0);
  int kt_pad = (ctile * 27 + 31) / 32 * 32;
  int kts = (KCH + ctile - 1) / ctile;
  auto wb = torch::zeros({ncol, (int64_t)kts * kt_pad}, w_flat2d.options());
  for (int kt = 0; kt < kts; ++kt) {
    int64_t k0 = (int64_t)kt * ctile * 27;
    int64_t klen = (int64_t)std::min(KCH - kt * ctile, ctile) * 27;
    wb.narrow(1, (int64_t)kt * kt_pad, klen).copy_(
        w_flat2d.narrow(1, k0, klen));
  }
  return wb;
}

static void launch_spatial(torch::Tensor in, torch::Tensor wb,
                           torch::Tensor out, SpDims sd, int stride,
                           int ctile = 0, const float* bn_ab = nullptr,
                           float* stats = nullptr) {
  int OWT = sd.TW % 32 == 0 ? 32 : (sd.TW % 16 == 0 ? 16 : 8);
  int chunk = stride == 1 ? 256 : 128;
  int ncolt = 32;
  // 64-col chunk-256 main instances (non-fused path): halves the slab
  // re-reads of the big stride-1 layers — measured 36.7 vs 37.5 ms/step
  // on the flagship (r2); the fused-stats form stays 32-col (a wide
  // variant of it mis-computed in an earlier build and is not needed:
  // the fused fwd runs a different instance family).
  if (stride == 1 && sd.NCOL >= 64 && ctile == 32) ncolt = 64;
  // chunk-512/CTILE-16 instances for thin-channel stride-1 layers
  // (L2-class): caller prepped wb with ctile=16 and passes ctile=165
  const bool c512 = (ctile == 165);
  // fused-stats wide form: numerically validated but measured SLOWER
  // (37.3 vs 36.7 ms/step — the stats reduction on 4 col-fragments adds
  // register pressure the fused fwd cannot afford); off by default.
  const bool wide_fused = getenv("COINN_WIDE256F") != nullptr;
  // small images (d8-class): 64-position chunks keep the grid dense;
  // wide-column instances halve the slab-staging redundancy there
  // (PMC: 86% SQ_WAIT on the 32-col chunk-64 forms)
  if (sd.TH * sd.TW < chunk) {
    chunk = 64;
    OWT = 8;  // the chunk-64 instances are OWT=8
    // swept on MI355X (r2): 128-col instances won at the deep layers
    // (37.7 vs 38.2 ms/step flagship) — slab re-reads drop 4x vs 32-col
    if (sd.NCOL >= 64 && ctile != 1) ncolt = sd.NCOL >= 128 ? 128 : 64;
  }
  int OHT = chunk / OWT;
  int wtiles = (sd.TW + OWT - 1) / OWT;
  int htiles = (sd.TH + OHT - 1) / OHT;
  int64_t nchunks = (int64_t)sd.N * sd.TD * htiles * wtiles;
  dim3 grid((unsigned)nchunks, (sd.NCOL + ncolt - 1) / ncolt);
  auto s = current_stream();
  const __bf16* ip = reinterpret_cast<const __bf16*>(in.data_ptr());
  const __bf16* wp = reinterpret_cast<const __bf16*>(wb.data_ptr());
  __bf16* op = reinterpret_cast<__bf16*>(out.data_ptr());
  // db kernel has no bn_ab parameter; the regular kernel always takes one
  // (hipLaunchKernelGGL cannot use default arguments)
  auto LDB = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, s, ip, wp, op, sd, nchunks);
  };
  auto L = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, s, ip, wp, op, sd, nchunks,
                       (const float*)nullptr, (float*)nullptr,
                       (const __bf16*)nullptr, (const float*)nullptr);
  };
  auto LF = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, s, ip, wp, op, sd, nchunks,
                       bn_ab, stats, (const __bf16*)nullptr,
                       (const float*)nullptr);
  };
  if (bn_ab != nullptr && stats != nullptr) {
    // fused normalize-on-load + epilogue-stats instances
    if (c512 && stride == 1) {
      int OHT5 = 512 / 32;
      int ht5 = (sd.TH + OHT5 - 1) / OHT5;
      int wt5 = (sd.TW + 31) / 32;
      nchunks = (int64_t)sd.N * sd.TD * ht5 * wt5;
      grid = dim3((unsigned)nchunks, (sd.NCOL + 31) / 32);
      LF(conv3d_spatial_kernel<32, 1, 16, 512, true, 1>);
      return;
    }
    if (stride == 1) {
      if (chunk == 64 && ncolt == 128)
        LF(conv3d_spatial_kernel<8, 1, 32, 64, true, 1, 128>);
      else if (chunk == 64 && ncolt == 64)
        LF(conv3d_spatial_kernel<8, 1, 32, 64, true, 1, 64>);
      else if (chunk == 64) LF(conv3d_spatial_kernel<8, 1, 32, 64, true, 1>);
      else if (OWT == 32 && wide_fused && ncolt == 64)
        LF(conv3d_spatial_kernel<32, 1, 32, 256, true, 1, 64>);
      else if (OWT == 32)
        LF(conv3d_spatial_kernel<32, 1, 32, 256, true, 1>);
      else if (OWT == 16 && wide_fused && ncolt == 64)
        LF(conv3d_spatial_kernel<16, 1, 32, 256, true, 1, 64>);
      else if (OWT == 16)
        LF(conv3d_spatial_kernel<16, 1, 32, 256, true, 1>);
      else LF(conv3d_spatial_kernel<8, 1, 32, 256, true, 1>);
    } else {
      if (chunk == 64 && ncolt == 128)
        LF(conv3d_spatial_kernel<8, 2, 16, 64, true, 1, 128>);
      else if (chunk == 64 && ncolt == 64)
        LF(conv3d_spatial_kernel<8, 2, 16, 64, true, 1, 64>);
      else if (chunk == 64) LF(conv3d_spatial_kernel<8, 2, 16, 64, true, 1>);
      else if (OWT == 32)
        LF(conv3d_spatial_kernel<32, 2, 16, 128, true, 1>);
      else if (OWT == 16)
        LF(conv3d_spatial_kernel<16, 2, 16, 128, true, 1>);
      else LF(conv3d_spatial_kernel<8, 2, 16, 128, true, 1>);
    }
    return;
  }
  if (bn_ab != nullptr) {
    // fused normalize-on-load instances (default tilings only)
    if (stride == 1) {
      if (chunk == 64 && ncolt == 128)
        LF(conv3d_spatial_kernel<8, 1, 32, 64, true, 0, 128>);
      else if (chunk == 64 && ncolt == 64)
        LF(conv3d_spatial_kernel<8, 1, 32, 64, true, 0, 64>);
      else if (chunk == 64) LF(conv3d_spatial_kernel<8, 1, 32, 64, true>);
      else if (OWT == 32) LF(conv3d_spatial_kernel<32, 1, 32, 256, true>);
      else if (OWT == 16) LF(conv3d_spatial_kernel<16, 1, 32, 256, true>);
      else LF(conv3d_spatial_kernel<8, 1, 32, 256, true>);
    } else {
      if (chunk == 64 && ncolt == 128)
        LF(conv3d_spatial_kernel<8, 2, 16, 64, true, 0, 128>);
      else if (chunk == 64 && ncolt == 64)
        LF(conv3d_spatial_kernel<8, 2, 16, 64, true, 0, 64>);
      else if (chunk == 64) LF(conv3d_spatial_kernel<8, 2, 16, 64, true>);
      else if (OWT == 32) LF(conv3d_spatial_kernel<32, 2, 16, 128, true>);
      else if (OWT == 16) LF(conv3d_spatial_kernel<16, 2, 16, 128, true>);
      else LF(conv3d_spatial_kernel<8, 2, 16, 128, true>);
    }
    return;
  }
  if (stride == 1 && chunk == 64 && ncolt == 128 && ctile == 32) {
    L(conv3d_spatial_kernel<8, 1, 32, 64, false, 0, 128>);
  } else if (stride == 2 && chunk == 64 && ncolt == 128) {
    L(conv3d_spatial_kernel<8, 2, 16, 64, false, 0, 128>);
  } else if (stride == 1 && chunk == 64 && ncolt == 64 && ctile == 32) {
    L(conv3d_spatial_kernel<8, 1, 32, 64, false, 0, 64>);
  } else if (stride == 2 && chunk == 64 && ncolt == 64) {
    L(conv3d_spatial_kernel<8, 2, 16, 64, false, 0, 64>);
  } else if (stride == 1 && ctile == 16) {
    // experimental double-buffered CTILE=16 instances (ctile_opt=16)
    if (chunk == 64) LDB(conv3d_spatial_db_kernel<8, 1, 16, 64>);
    else if (OWT == 32) LDB(conv3d_spatial_db_kernel<32, 1, 16>);
    else if (OWT == 16) LDB(conv3d_spatial_db_kernel<16, 1, 16>);
    else LDB(conv3d_spatial_db_kernel<8, 1, 16>);
  } else if (stride == 1 && ctile == 1) {
    // single-channel (first-layer) instances: one 32-k-step covers the
    // whole 27-tap K, slab is [1][3][H2][W2]
    if (chunk == 64) L(conv3d_spatial_kernel<8, 1, 1, 64>);
    else if (OWT == 32) L(conv3d_spatial_kernel<32, 1, 1>);
    else if (OWT == 16) L(conv3d_spatial_kernel<16, 1, 1>);
    else L(conv3d_spatial_kernel<8, 1, 1>);
  } else if (c512 && stride == 1) {
    // grid must be recomputed for the larger chunk
    {
      int OHT5 = 512 / 32;
      int ht5 = (sd.TH + OHT5 - 1) / OHT5;
      int wt5 = (sd.TW + 31) / 32;
      nchunks = (int64_t)sd.N * sd.TD * ht5 * wt5;
      grid = dim3((unsigned)nchunks, (sd.NCOL + 31) / 32);
    }
    L(conv3d_spatial_kernel<32, 1, 16, 512>);
  } else if (stride == 1 && ncolt == 64 && chunk == 256) {
    if (OWT == 32) L(conv3d_spatial_kernel<32, 1, 32, 256, false, 0, 64>);
    else if (OWT == 16)
      L(conv3d_spatial_kernel<16, 1, 32, 256, false, 0, 64>);
    else L(conv3d_spatial_kernel<8, 1, 32, 256, false, 0, 64>);
  } else if (stride == 1) {
    if (chunk == 64) L(conv3d_spatial_kernel<8, 1, 32, 64>);
    else if (OWT == 32) L(conv3d_spatial_kernel<32, 1, 32>);
    else if (OWT == 16) L(conv3d_spatial_kernel<16, 1, 32>);
    else L(conv3d_spatial_kernel<8, 1, 32>);
  } else {
    if (chunk == 64) L(conv3d_spatial_kernel<8, 2, 16, 64>);
    else if (OWT == 32) L(conv3d_spatial_kernel<32, 2, 16>);
    else if (OWT == 16) L(conv3d_spatial_kernel<16, 2, 16>);
    else L(conv3d_spatial_kernel<8, 2, 16>);
  }
}

torch::Tensor conv3d_fwd_spatial(torch::Tensor x, torch::Tensor w,
                                 int64_t stride, int64_t ctile_opt,
                                 c10::optional<torch::Tensor> bn_ab_opt) {
  torch::Tensor bn_ab = bn_ab_opt.value_or(torch::Tensor());
  CHECK_GPU(x);
  auto xc = x.contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  TORCH_CHECK(xc.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(stride == 1 || stride == 2);
  const bool fuse = bn_ab.defined() && bn_ab.numel() > 0;
  torch::Tensor ab;
  if (fuse) {
    TORCH_CHECK(ctile_opt == 0, "fused BN requires default tiling");
    ab = bn_ab.to(torch::kFloat32).contiguous();
    TORCH_CHECK(ab.numel() == 2 * x.size(1), "bn_ab must be [Cin,2]");
  }
  SpDims sd;
  sd.N = (int)xc.size(0); sd.KCH = (int)xc.size(1);
  sd.D = (int)xc.size(2); sd.H = (int)xc.size(3); sd.W = (int)xc.size(4);
  sd.NCOL = (int)wc.size(0);
  sd.TD = (sd.D + 2 - 3) / (int)stride + 1;
  sd.TH = (sd.H + 2 - 3) / (int)stride + 1;
  sd.TW = (sd.W + 2 - 3) / (int)stride + 1;
  // ctile_opt=1 opts into the CTILE=1 single-channel instances;
  // ctile_opt=16 into the double-buffered CTILE=16 instances
  // (both stride 1 only); 0 = the default tiling.
  int ctile = stride == 1
                  ? ((ctile_opt == 1 || ctile_opt == 16) ? (int)ctile_opt : 32)
                  : 16;
  auto wb = prep_wb(wc.reshape({sd.NCOL, (int64_t)sd.KCH * 27}), sd.KCH,
                    ctile);
  sd.Kpad = (int)wb.size(1);
  auto out = torch::empty({sd.N, sd.NCOL, sd.TD, sd.TH, sd.TW},
                          xc.options());
  launch_spatial(xc, wb, out, sd, (int)stride, ctile,
                 fuse ? ab.data_ptr<float>() : nullptr);
  return out;
}

// fused fwd + epilogue BN statistics: returns {out, stats[64][Cout][2]}
// (hash-sliced partial sums; callers fold with stats.sum(0)).
std::vector<torch::Tensor> conv3d_fwd_spatial_stats(
    torch::Tensor x, torch::Tensor w, int64_t stride,
    c10::optional<torch::Tensor> bn_ab_opt) {
  torch::Tensor bn_ab = bn_ab_opt.value_or(torch::Tensor());
  CHECK_GPU(x);
  auto xc = x.contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  TORCH_CHECK(xc.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(stride == 1 || stride == 2);
  TORCH_CHECK(bn_ab.defined() && bn_ab.numel() > 0,
              "stats form requires the fused-BN path");
  auto ab = bn_ab.to(torch::kFloat32).contiguous();
  SpDims sd;
  sd.N = (int)xc.size(0); sd.KCH = (int)xc.size(1);
  sd.D = (int)xc.size(2); sd.H = (int)xc.size(3); sd.W = (int)xc.size(4);
  sd.NCOL = (int)wc.size(0);
  sd.TD = (sd.D + 2 - 3) / (int)stride + 1;
  sd.TH = (sd.H + 2 - 3) / (int)stride + 1;
  sd.TW = (sd.W + 2 - 3) / (int)stride + 1;
  int ctile = stride == 1 ? 32 : 16;
  // chunk-512 thin-channel fused fwd: measured 35.8 vs 36.1 ms/step
  // (r2); COINN_C512F=0 reverts
  const char* c5e = getenv("COINN_C512F");
  const bool c512 = (!c5e || c5e[0] != '0') && stride == 1 &&
                    sd.KCH <= 32 && sd.TW % 32 == 0 &&
                    sd.TH * sd.TW >= 512;
  if (c512) ctile = 16;
  auto wb = prep_wb(wc.reshape({sd.NCOL, (int64_t)sd.KCH * 27}), sd.KCH,
                    ctile);
  sd.Kpad = (int)wb.size(1);
  auto out = torch::empty({sd.N, sd.NCOL, sd.TD, sd.TH, sd.TW},
                          xc.options());
  auto stats = torch::zeros({64, sd.NCOL, 2},
                            xc.options().dtype(torch::kFloat32));
  launch_spatial(xc, wb, out, sd, (int)stride, c512 ? 165 : ctile,
                 ab.data_ptr<float>(), stats.data_ptr<float>());
  return {out, stats};
}

torch::Tensor conv3d_dgrad_spatial(torch::Tensor go, torch::Tensor w,
                                   std::vector<int64_t> in_shape) {
  CHECK_GPU(go);
  auto g = go.to(torch::kBFloat16).contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  SpDims sd;
  sd.N = (int)g.size(0); sd.KCH = (int)g.size(1);
  sd.D = (int)g.size(2); sd.H = (int)g.size(3); sd.W = (int)g.size(4);
  sd.NCOL = (int)in_shape[1];
  sd.TD = (int)in_shape[2]; sd.TH = (int)in_shape[3];
  sd.TW = (int)in_shape[4];
  int Cout = (int)wc.size(0), Cin = (int)wc.size(1);
  auto wf = wc.reshape({Cout, Cin, 27}).flip(-1).permute({1, 0, 2})
                .reshape({Cin, (int64_t)Cout * 27}).contiguous();
  // thin-channel large-plane layers: the chunk-512/CTILE-16 instance
  // doubles the m amortizing each staged slab (measured 36.2 vs 36.6
  // ms/step on the flagship, r2; COINN_C512=0 reverts)
  const char* c512_env = getenv("COINN_C512");
  const bool c512 = (!c512_env || c512_env[0] != '0') && sd.KCH <= 32 &&
                    sd.TW % 32 == 0 && sd.TH * sd.TW >= 512;
  auto wb = prep_wb(wf, Cout, c512 ? 16 : 32);
  sd.Kpad = (int)wb.size(1);
  auto dx = torch::empty(in_shape, g.options());
  launch_spatial(g, wb, dx, sd, 1, c512 ? 165 : 0);
  return dx;
}

// stride-1 dgrad that ALSO folds the BN-backward reduction of the dz it
// writes: returns {dz, bsums[64][Cin][2]} with bsums = hash-sliced
// partials of (sum dz*mask, sum dz*xhat*mask). bn_prm packs
// [mean, rstd, gamma, beta] per channel of the BN being backpropped.
std::vector<torch::Tensor> conv3d_dgrad_spatial_bnbwd(
    torch::Tensor go, torch::Tensor w, std::vector<int64_t> in_shape,
    torch::Tensor xraw, torch::Tensor bn_prm) {
  CHECK_GPU(go);
  auto g = go.to(torch::kBFloat16).contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  auto xr = xraw.contiguous();
  TORCH_CHECK(xr.scalar_type() == torch::kBFloat16, "xraw must be bf16");
  auto prm = bn_prm.to(torch::kFloat32).contiguous();
  SpDims sd;
  sd.N = (int)g.size(0); sd.KCH = (int)g.size(1);
  sd.D = (int)g.size(2); sd.H = (int)g.size(3); sd.W = (int)g.size(4);
  sd.NCOL = (int)in_shape[1];
  sd.TD = (int)in_shape[2]; sd.TH = (int)in_shape[3];
  sd.TW = (int)in_shape[4];
  TORCH_CHECK(prm.numel() == 4 * sd.NCOL, "bn_prm must be [Cin,4]");
  int Cout = (int)wc.size(0), Cin = (int)wc.size(1);
  auto wf = wc.reshape({Cout, Cin, 27}).flip(-1).permute({1, 0, 2})
                .reshape({Cin, (int64_t)Cout * 27}).contiguous();
  auto wb = prep_wb(wf, Cout, 32);
  sd.Kpad = (int)wb.size(1);
  auto dx = torch::empty(in_shape, g.options());
  auto bsums = torch::zeros({64, sd.NCOL, 2},
                            g.options().dtype(torch::kFloat32));

  int OWT = sd.TW % 32 == 0 ? 32 : (sd.TW % 16 == 0 ? 16 : 8);
  int chunk = 256;
  if (sd.TH * sd.TW < chunk) { chunk = 64; OWT = 8; }
  int OHT = chunk / OWT;
  int wtiles = (sd.TW + OWT - 1) / OWT;
  int htiles = (sd.TH + OHT - 1) / OHT;
  int64_t nchunks = (int64_t)sd.N * sd.TD * htiles * wtiles;
  dim3 grid((unsigned)nchunks, (sd.NCOL + 31) / 32);
  auto st = current_stream();
  auto LB = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, st,
                       reinterpret_cast<const __bf16*>(g.data_ptr()),
                       reinterpret_cast<const __bf16*>(wb.data_ptr()),
                       reinterpret_cast<__bf16*>(dx.data_ptr()), sd, nchunks,
                       (const float*)nullptr, bsums.data_ptr<float>(),
                       reinterpret_cast<const __bf16*>(xr.data_ptr()),
                       prm.data_ptr<float>());
  };
  if (chunk == 64) LB(conv3d_spatial_kernel<8, 1, 32, 64, false, 2>);
  else if (OWT == 32) LB(conv3d_spatial_kernel<32, 1, 32, 256, false, 2>);
  else if (OWT == 16) LB(conv3d_spatial_kernel<16, 1, 32, 256, false, 2>);
  else LB(conv3d_spatial_kernel<8, 1, 32, 256, false, 2>);
  return {dx, bsums};
}

// ---------------------------------------------------------------------------
// Stride-2 DGRAD, parity classes through the spatial-slab structure.
// dx decomposes by (id,ih,iw) mod 2 into 8 dense sub-problems; within a
// class the taps shift the go window by only {0,+1} per axis, so a block
// stages a TINY go slab [co32][2][OHT+1][OWT+1] once per 32-co tile and
// computes K = 32*T tap-products from it (T = 1..8 taps per class). The
// per-class weight matrix WBc[ci][co*T + r] is prepared host-side
// (tap-gathered), contiguous 16B per lane. blockIdx.z = class.
// ---------------------------------------------------------------------------
template <int OWT>
__global__ __launch_bounds__(256) void conv3d_dgrad_s2_sp_kernel(
    const __bf16* __restrict__ go, const __bf16* __restrict__ wb,
    __bf16* __restrict__ dx, SpDims sd, int cls_off0, int cls_off1,
    int cls_off2, int cls_off3, int cls_off4, int cls_off5, int cls_off6,
    int cls_off7) {
  constexpr int OHT = 128 / OWT;
  constexpr int W2 = OWT + 4;        // OWT+1 used; padded
  constexpr int H2 = OHT + 1;
  constexpr int MPW = (OWT * OHT) / 64;  // = 2
  __shared__ __bf16 sG[32][2][H2][W2];
  __shared__ unsigned short sKtab[32 * 8 + 8];

  const int cls = blockIdx.z;
  const int a = (cls >> 2) & 1, b = (cls >> 1) & 1, c = cls & 1;
  const int l2w = c, l2h = b;
  const int T = 1 << (a + b + c);
  const int cls_offs[8] = {cls_off0, cls_off1, cls_off2, cls_off3,
                           cls_off4, cls_off5, cls_off6, cls_off7};
  const int wb_cls = cls_offs[cls];

  // dx sub-grid dims for this class
  const int Da = (sd.TD - a + 1) >> 1;
  const int Hb = (sd.TH - b + 1) >> 1;
  const int Wc = (sd.TW - c + 1) >> 1;

  const int ncol0 = blockIdx.y * 32;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int row = lane & 15, kg = lane >> 4;

  const int wtiles = (Wc + OWT - 1) / OWT;
  const int htiles = (Hb + OHT - 1) / OHT;
  const int64_t nchunks = (int64_t)sd.N * Da * htiles * wtiles;
  if (blockIdx.x >= nchunks) return;

  int64_t t = blockIdx.x;
  const int wt = (int)(t % wtiles);
  t /= wtiles;
  const int ht = (int)(t % htiles);
  t /= htiles;
  const int tdp = (int)(t % Da);     // id' (sub-grid d)
  const int n = (int)(t / Da);
  const int oh0 = ht * OHT, ow0 = wt * OWT;

  // k -> slab offset: k = co_l*T + r; r -> (dod, doh, dow) in {0,1}
  for (int k = tid; k < 32 * T; k += 256) {
    const int co_l = k >> (a + b + c);
    const int r = k & (T - 1);
    const int tw_i = r & ((1 << l2w) - 1);
    const int th_i = (r >> l2w) & ((1 << l2h) - 1);
    const int td_i = r >> (l2w + l2h);
    const int dod = a ? (1 - td_i) : 0;
    const int doh = b ? (1 - th_i) : 0;
    const int dow = c ? (1 - tw_i) : 0;
    sKtab[k] = (unsigned short)(((co_l * 2 + dod) * H2 + doh) * W2 + dow);
  }

  f32x4 acc[MPW][2];
#pragma unroll
  for (int i = 0; i < MPW; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int64_t OHW = (int64_t)sd.H * sd.W;  // go plane (input-side dims)
  const int64_t go_n = (int64_t)n * sd.KCH * sd.D * OHW;
  const int kts = (sd.KCH + 31) / 32;        // co tiles
  const int KT = 32 * T;                     // k per co tile (mult of 32)

  for (int kt = 0; kt < kts; ++kt) {
    constexpr int NRMAX = 32 * 2 * H2;
    if (kt) __syncthreads();
    else __syncthreads();  // ktab ready
    for (int r = tid; r < NRMAX; r += 256) {
      const int hrow = r % H2;
      const int p = (r / H2) & 1;
      const int co = r / (2 * H2);
      const int od = tdp + p;                // dod in {0,1}
      const int oh = oh0 + hrow;
      const int ch = kt * 32 + co;
      __bf16* dst = &sG[co][p][hrow][0];
      const bool row_ok = od < sd.D && oh < sd.H && ch < sd.KCH &&
                          (a || p == 0);     // a==0 uses only plane 0
      if (!row_ok) {
#pragma unroll
        for (int col = 0; col < W2; ++col) dst[col] = (__bf16)0.f;
        continue;
      }
      const __bf16* src = go + go_n + ((int64_t)ch * sd.D + od) * OHW +
                          (int64_t)oh * sd.W + ow0;
#pragma unroll
      for (int v = 0; v < OWT / 8; ++v) {
        if (ow0 + v * 8 + 7 < sd.W) {
          bf16x8 vec = *reinterpret_cast<const bf16x8*>(src + v * 8);
#pragma unroll
          for (int j = 0; j < 8; ++j) dst[v * 8 + j] = vec[j];
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int ow = ow0 + v * 8 + j;
            dst[v * 8 + j] = (ow < sd.W) ? src[v * 8 + j] : (__bf16)0.f;
          }
        }
      }
      {  // +1 tail column
        const int ow = ow0 + OWT;
        dst[OWT] = (ow < sd.W) ? src[OWT] : (__bf16)0.f;
      }
    }
    __syncthreads();

#pragma unroll 1
    for (int ks = 0; ks < KT / 32; ++ks) {
      bf16x8 afrag[MPW];
      {
        const int kb = ks * 32 + kg * 8;
        const u16x8 kt8 = *reinterpret_cast<const u16x8*>(&sKtab[kb]);
        const __bf16* slab = &sG[0][0][0][0];
#pragma unroll
        for (int i = 0; i < MPW; ++i) {
          const int m = (wave * MPW + i) * 16 + row;
          const int base = (m / OWT) * W2 + (m % OWT);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            afrag[i][j] = slab[base + kt8[j]];
        }
      }
      bf16x8 bfrag[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int col = ncol0 + i * 16 + row;
        const int64_t off = (int64_t)col * sd.Kpad + wb_cls + kt * KT +
                            ks * 32 + kg * 8;
        bfrag[i] = (col < sd.NCOL)
                       ? *reinterpret_cast<const bf16x8*>(wb + off)
                       : bf16x8{};
      }
#pragma unroll
      for (int i = 0; i < MPW; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
  }

  // epilogue: dx[n][ci][2*id'+a][2*ih'+b][2*iw'+c]
  const int64_t THW = (int64_t)sd.TH * sd.TW;
  const int64_t dx_n = (int64_t)n * sd.NCOL * sd.TD * THW;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < MPW; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int col = ncol0 + j * 16 + ccol;
      if (col >= sd.NCOL) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = (wave * MPW + i) * 16 + crow0 + r;
        const int ihp = oh0 + m / OWT;
        const int iwp = ow0 + m % OWT;
        if (ihp < Hb && iwp < Wc)
          dx[dx_n + ((int64_t)col * sd.TD + (2 * tdp + a)) * THW +
             (int64_t)(2 * ihp + b) * sd.TW + (2 * iwp + c)] =
              (__bf16)(acc[i][j][r]);
      }
    }
  }
}

torch::Tensor conv3d_dgrad_s2_spatial(torch::Tensor go, torch::Tensor w,
                                      std::vector<int64_t> in_shape) {
  CHECK_GPU(go);
  auto g = go.to(torch::kBFloat16).contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  int Cout = (int)wc.size(0), Cin = (int)wc.size(1);
  SpDims sd;
  sd.N = (int)g.size(0); sd.KCH = Cout;
  sd.D = (int)g.size(2); sd.H = (int)g.size(3); sd.W = (int)g.size(4);
  sd.NCOL = Cin;
  sd.TD = (int)in_shape[2]; sd.TH = (int)in_shape[3];
  sd.TW = (int)in_shape[4];

  // per-class tap-gathered weights: WBc[ci][co*T + r]. The tap index
  // tensors are compile-time constants — cached per device so no H2D
  // copy happens on the hot path (pageable H2D is illegal inside a
  // hipGraph capture).
  auto w3 = wc.reshape({Cout, Cin, 27}).permute({1, 0, 2}).contiguous();
  static std::vector<torch::Tensor> s_idx;       // [8], device-resident
  static torch::Device s_dev = torch::kCPU;
  if (s_idx.empty() || s_dev != wc.device()) {
    s_idx.clear();
    for (int cls = 0; cls < 8; ++cls) {
      int a = (cls >> 2) & 1, b = (cls >> 1) & 1, c = cls & 1;
      std::vector<int64_t> taps;
      for (int td_i = 0; td_i < (a ? 2 : 1); ++td_i)
        for (int th_i = 0; th_i < (b ? 2 : 1); ++th_i)
          for (int tw_i = 0; tw_i < (c ? 2 : 1); ++tw_i) {
            int kd = a ? td_i * 2 : 1;
            int kh = b ? th_i * 2 : 1;
            int kw = c ? tw_i * 2 : 1;
            taps.push_back(kd * 9 + kh * 3 + kw);
          }
      s_idx.push_back(torch::tensor(taps, torch::TensorOptions()
                                              .dtype(torch::kLong))
                          .to(wc.device()));
    }
    s_dev = wc.device();
  }
  std::vector<torch::Tensor> blocks;
  std::vector<int64_t> offs(8, 0);
  int64_t cur = 0;
  for (int cls = 0; cls < 8; ++cls) {
    auto blk = w3.index_select(2, s_idx[cls]).reshape({Cin, -1});
    offs[cls] = cur;
    cur += blk.size(1);
    blocks.push_back(blk);
  }
  auto wb = torch::cat(blocks, 1).contiguous();  // [Cin][Cout*27]
  sd.Kpad = (int)wb.size(1);

  auto dx = torch::empty(in_shape, g.options());
  int Wc = (sd.TW + 1) >> 1, Hb = (sd.TH + 1) >> 1, Da = (sd.TD + 1) >> 1;
  int OWT = Wc % 32 == 0 ? 32 : (Wc % 16 == 0 ? 16 : 8);
  int OHT = 128 / OWT;
  int wtiles = (Wc + OWT - 1) / OWT;
  int htiles = (Hb + OHT - 1) / OHT;
  int64_t nchunks = (int64_t)sd.N * Da * htiles * wtiles;
  dim3 grid((unsigned)nchunks, (Cin + 31) / 32, 8);
  auto s = current_stream();
  auto L = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, s,
                       reinterpret_cast<const __bf16*>(g.data_ptr()),
                       reinterpret_cast<const __bf16*>(wb.data_ptr()),
                       reinterpret_cast<__bf16*>(dx.data_ptr()), sd,
                       (int)offs[0], (int)offs[1], (int)offs[2],
                       (int)offs[3], (int)offs[4], (int)offs[5],
                       (int)offs[6], (int)offs[7]);
  };
  if (OWT == 32) L(conv3d_dgrad_s2_sp_kernel<32>);
  else if (OWT == 16) L(conv3d_dgrad_s2_sp_kernel<16>);
  else L(conv3d_dgrad_s2_sp_kernel<8>);
  return dx;
}
