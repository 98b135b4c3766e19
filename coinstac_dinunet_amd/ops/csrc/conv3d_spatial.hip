// Stride-1 Conv3d fwd/dgrad with spatial-slab tap reuse.
// The igemm formulation re-gathers x once per (ci,tap) — 27x traffic.
// Here a block stages one input spatial slab [kch32][3][OHT+2][OWT+4]
// per 32-channel tile and computes the whole K loop from it: per k-step
// the A fragments read the slab at tap-shifted offsets (scalar LDS u16;
// the slab is the reuse win) and the B fragments are contiguous 16-byte
// global loads from the small L2-resident prepared-weight matrix
// WB[ncol][Kpad] (fwd: w reshaped; dgrad: w tap-flipped + transposed so
// the SAME gather geometry serves both directions).
//
// Geometry: out tile = 256 m x 32 ncol; 4 waves each own 4 m-fragments
// x 2 ncol-fragments (acc 8 x f32x4); chunk = (n, d, h-tile, w-tile).
#include "common.h"

#include <hip/hip_bf16.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;

struct SpDims {
  int N, KCH;        // input-side channels (fwd: Cin; dgrad: Cout)
  int D, H, W;       // input-side spatial (fwd: x dims; dgrad: go dims)
  int NCOL;          // output-side channels
  int TD, TH, TW;    // output-side spatial (== input-side for stride 1)
  int Kpad;          // padded row length of WB
};

#ifndef LDA_PAD
#define LDA_PAD 8
#endif

template <int OWT>
__global__ __launch_bounds__(256) void conv3d_s1_spatial_kernel(
    const __bf16* __restrict__ in, const __bf16* __restrict__ wb,
    __bf16* __restrict__ out, SpDims sd, int64_t nchunks) {
  constexpr int OHT = 256 / OWT;
  constexpr int W2 = OWT + 4;
  constexpr int H2 = OHT + 2;
  __shared__ __bf16 sX[32][3][H2][W2];
  // per-k element offsets into the slab (tap decode hoisted off the hot
  // path: the A-fragment read becomes base(m) + sKtab[k])
  __shared__ unsigned short sKtab[32 * 27 + 8];

  const int ncol0 = blockIdx.y * 32;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int row = lane & 15, kg = lane >> 4;

  const int wtiles = (sd.TW + OWT - 1) / OWT;
  const int htiles = (sd.TH + OHT - 1) / OHT;

  const int64_t z = blockIdx.x;
  if (z >= nchunks) return;
  int64_t t = z;
  const int wt = (int)(t % wtiles);
  t /= wtiles;
  const int ht = (int)(t % htiles);
  t /= htiles;
  const int td = (int)(t % sd.TD);
  const int n = (int)(t / sd.TD);
  const int oh0 = ht * OHT, ow0 = wt * OWT;

  // build the k -> slab-offset table once per block
  for (int k = tid; k < 32 * 27; k += 256) {
    const int cl = k / 27;
    const int r27 = k - cl * 27;
    const int a = r27 / 9, b = (r27 / 3) % 3, c = r27 % 3;
    sKtab[k] = (unsigned short)(((cl * 3 + a) * H2 + b) * W2 + c);
  }

  // wave owns m-fragments wave*4 .. wave*4+3 (16 m each)
  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int64_t HW = (int64_t)sd.H * sd.W;
  const int64_t in_n = (int64_t)n * sd.KCH * sd.D * HW;
  const int kts = (sd.KCH + 31) / 32;

  for (int kt = 0; kt < kts; ++kt) {
    // ---- stage the input slab for channels [kt*32, kt*32+32) ----------
    // row-wise: each thread owns whole (ch, kd, h) rows; the OWT interior
    // columns are IN-BOUNDS by construction (OWT divides W, stride 1), so
    // they load as OWT/8 16-byte vectors; only the 4 halo columns are
    // bounds-checked scalars. (The elementwise form was 135 scalar
    // loads/thread and made the kernel VALU-bound.)
    constexpr int NROWS = 32 * 3 * H2;
    if (kt) __syncthreads();
    for (int r = tid; r < NROWS; r += 256) {
      const int hrow = r % H2;
      const int a = (r / H2) % 3;
      const int c = r / (3 * H2);
      const int id = td - 1 + a;
      const int ih = oh0 - 1 + hrow;
      const int ch = kt * 32 + c;
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
      // left halo (iw = ow0-1)
      dst[0] = (ow0 > 0) ? src[ow0 - 1] : (__bf16)0.f;
      // interior: iw = ow0 .. ow0+OWT-1 (aligned 16B when ow0%8==0)
#pragma unroll
      for (int v = 0; v < OWT / 8; ++v) {
        bf16x8 vec = *reinterpret_cast<const bf16x8*>(src + ow0 + v * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) dst[1 + v * 8 + j] = vec[j];
      }
      // right halo (iw = ow0+OWT .. ow0+OWT+2)
#pragma unroll
      for (int e = 0; e < 3; ++e) {
        const int iw = ow0 + OWT + e;
        dst[1 + OWT + e] = (iw < sd.W) ? src[iw] : (__bf16)0.f;
      }
    }
    __syncthreads();

    // ---- 27 k-steps of 32 over (ch_local, tap) ------------------------
    const int kbase_g = kt * 32 * 27;
#pragma unroll 1
    for (int ks = 0; ks < 27; ++ks) {
      // A fragments: one per m-frag; element (kg,j): k = ks*32+kg*8+j;
      // addresses come from the precomputed table (one 16B LDS read)
      bf16x8 afrag[4];
      {
        const int kb = ks * 32 + kg * 8;
        const u16x8 kt = *reinterpret_cast<const u16x8*>(&sKtab[kb]);
        const __bf16* slab = &sX[0][0][0][0];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int m = (wave * 4 + i) * 16 + row;
          const int base = (m / OWT) * W2 + (m % OWT);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            afrag[i][j] = slab[base + kt[j]];
        }
      }
      // B fragments: wb[ncol][Kpad], contiguous 16B per lane (L2)
      bf16x8 bfrag[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int col = ncol0 + i * 16 + row;
        const int64_t off =
            (int64_t)col * sd.Kpad + kbase_g + ks * 32 + kg * 8;
        bfrag[i] = (col < sd.NCOL)
                       ? *reinterpret_cast<const bf16x8*>(wb + off)
                       : bf16x8{};
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
  }

  // ---- epilogue: out[n][col][td][oh0+][ow0+] --------------------------
  const int64_t THW = (int64_t)sd.TH * sd.TW;
  const int64_t out_n = (int64_t)n * sd.NCOL * sd.TD * THW;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int col = ncol0 + j * 16 + ccol;
      if (col >= sd.NCOL) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = (wave * 4 + i) * 16 + crow0 + r;
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
static torch::Tensor prep_wb_fwd(torch::Tensor w) {
  // WB[Cout][Cin*27] padded to 8-elem rows
  int Cout = (int)w.size(0);
  int K = (int)(w.numel() / Cout);
  int Kpad = (K + 7) / 8 * 8;
  auto wb = torch::zeros({Cout, Kpad}, w.options());
  wb.narrow(1, 0, K).copy_(w.reshape({Cout, K}));
  return wb;
}

static torch::Tensor prep_wb_dgrad(torch::Tensor w) {
  // WB[Cin][Cout*27] with taps flipped: wb[ci][co*27+r] = w[co][ci][26-r]
  int Cout = (int)w.size(0), Cin = (int)w.size(1);
  auto wf = w.reshape({Cout, Cin, 27}).flip(-1).permute({1, 0, 2})
                .reshape({Cin, Cout * 27});
  int K = Cout * 27;
  int Kpad = (K + 7) / 8 * 8;
  auto wb = torch::zeros({Cin, Kpad}, w.options());
  wb.narrow(1, 0, K).copy_(wf);
  return wb;
}

static void launch_spatial(torch::Tensor in, torch::Tensor wb,
                           torch::Tensor out, SpDims sd) {
  int OWT = sd.TW % 32 == 0 ? 32 : (sd.TW % 16 == 0 ? 16 : 8);
  int OHT = 256 / OWT;
  int wtiles = (sd.TW + OWT - 1) / OWT;
  int htiles = (sd.TH + OHT - 1) / OHT;
  int64_t nchunks = (int64_t)sd.N * sd.TD * htiles * wtiles;
  dim3 grid((unsigned)nchunks, (sd.NCOL + 31) / 32);
  auto s = current_stream();
  const __bf16* ip = reinterpret_cast<const __bf16*>(in.data_ptr());
  const __bf16* wp = reinterpret_cast<const __bf16*>(wb.data_ptr());
  __bf16* op = reinterpret_cast<__bf16*>(out.data_ptr());
  if (OWT == 32)
    hipLaunchKernelGGL(conv3d_s1_spatial_kernel<32>, grid, dim3(256), 0, s,
                       ip, wp, op, sd, nchunks);
  else if (OWT == 16)
    hipLaunchKernelGGL(conv3d_s1_spatial_kernel<16>, grid, dim3(256), 0, s,
                       ip, wp, op, sd, nchunks);
  else
    hipLaunchKernelGGL(conv3d_s1_spatial_kernel<8>, grid, dim3(256), 0, s,
                       ip, wp, op, sd, nchunks);
}

torch::Tensor conv3d_fwd_spatial(torch::Tensor x, torch::Tensor w) {
  CHECK_GPU(x);
  auto xc = x.contiguous();
  auto wc = w.to(torch::kBFloat16).contiguous();
  TORCH_CHECK(xc.scalar_type() == torch::kBFloat16);
  SpDims sd;
  sd.N = (int)xc.size(0); sd.KCH = (int)xc.size(1);
  sd.D = (int)xc.size(2); sd.H = (int)xc.size(3); sd.W = (int)xc.size(4);
  sd.NCOL = (int)wc.size(0);
  sd.TD = sd.D; sd.TH = sd.H; sd.TW = sd.W;
  auto wb = prep_wb_fwd(wc);
  sd.Kpad = (int)wb.size(1);
  auto out = torch::empty({sd.N, sd.NCOL, sd.TD, sd.TH, sd.TW},
                          xc.options());
  launch_spatial(xc, wb, out, sd);
  return out;
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
  auto wb = prep_wb_dgrad(wc);
  sd.Kpad = (int)wb.size(1);
  auto dx = torch::empty(in_shape, g.options());
  launch_spatial(g, wb, dx, sd);
  return dx;
}
