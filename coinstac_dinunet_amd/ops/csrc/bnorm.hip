// Fused BatchNorm3d(+ReLU) training fwd/bwd for NCDHW bf16.
// MIOpen's spatial BN runs ~0.8 TB/s on these shapes (profiles/r01);
// these kernels are plain HBM-roofline streaming: grid-stride fp32
// reductions (vectorized 8x bf16 loads, per-wave shuffle + 2 fp32
// atomics per block) and an elementwise normalize pass with the ReLU
// folded in. Backward uses the saved INPUT x: it recomputes
// z = gamma*xhat + beta for the ReLU mask (y == relu(z)), so the fused
// ReLU costs no extra saved tensor.
#include "common.h"

#include <hip/hip_bf16.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;

__device__ inline void block_reduce_2(float s, float sx, float* out0,
                                      float* out1) {
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
    s += __shfl_down(s, off);
    sx += __shfl_down(sx, off);
  }
  __shared__ float ps[8], pss[8];
  const int wave = threadIdx.x / WAVE_SIZE;
  if ((threadIdx.x & 63) == 0) { ps[wave] = s; pss[wave] = sx; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float ts = 0.f, tsx = 0.f;
    for (int i = 0; i < (int)(blockDim.x / WAVE_SIZE); ++i) {
      ts += ps[i]; tsx += pss[i];
    }
    atomicAdd(out0, ts);
    atomicAdd(out1, tsx);
  }
}

// ---- fwd pass 1: per-channel sum / sumsq (grid.y = n*c) -------------------
__global__ void bn_reduce_kernel(const __bf16* __restrict__ x,
                                 float* __restrict__ sums,  // [C][2]
                                 int C, int64_t spatial) {
  const int nc = blockIdx.y;
  const int c = nc % C;
  const __bf16* xp = x + (int64_t)nc * spatial;
  float s = 0.f, ss = 0.f;
  int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (; i + 8 <= spatial; i += stride) {
    bf16x8v v = *reinterpret_cast<const bf16x8v*>(xp + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)v[j];
      s += f;
      ss += f * f;
    }
  }
  if (i < spatial)
    for (int64_t j = i; j < spatial; ++j) {
      float f = (float)xp[j];
      s += f;
      ss += f * f;
    }
  block_reduce_2(s, ss, &sums[c * 2], &sums[c * 2 + 1]);
}

// ---- fwd pass 2: y = relu((x - mean) * rstd * gamma + beta [+ res]) -------
// RES: residual branch added before the ReLU — fuses ResNet's
// bn2 -> (+identity) -> ReLU into the one normalize pass.
template <bool RELU, bool RES = false>
__global__ void bn_normalize_kernel(const __bf16* __restrict__ x,
                                    __bf16* __restrict__ y,
                                    const float* __restrict__ mean_rstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ beta, int C,
                                    int64_t spatial,
                                    const __bf16* __restrict__ res = nullptr) {
  const int nc = blockIdx.y;
  const int c = nc % C;
  const float mean = mean_rstd[c * 2], rstd = mean_rstd[c * 2 + 1];
  const float g = gamma[c] * rstd, b = beta[c] - mean * g;
  const __bf16* xp = x + (int64_t)nc * spatial;
  const __bf16* rp = RES ? res + (int64_t)nc * spatial : nullptr;
  __bf16* yp = y + (int64_t)nc * spatial;
  int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (; i + 8 <= spatial; i += stride) {
    bf16x8v v = *reinterpret_cast<const bf16x8v*>(xp + i);
    bf16x8v rv;
    if (RES) rv = *reinterpret_cast<const bf16x8v*>(rp + i);
    bf16x8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)v[j] * g + b;
      if (RES) f += (float)rv[j];
      if (RELU) f = fmaxf(f, 0.f);
      o[j] = (__bf16)f;
    }
    *reinterpret_cast<bf16x8v*>(yp + i) = o;
  }
  if (i < spatial)
    for (int64_t j = i; j < spatial; ++j) {
      float f = (float)xp[j] * g + b;
      if (RES) f += (float)rp[j];
      if (RELU) f = fmaxf(f, 0.f);
      yp[j] = (__bf16)f;
    }
}

// ---- bwd pass 1: per-channel sum(dz), sum(dz * xhat) ----------------------
// RES: the fused residual path — the ReLU mask is (affine(x)+res) > 0.
template <bool RELU, bool RES = false>
__global__ void bn_bwd_reduce_kernel(const __bf16* __restrict__ dy,
                                     const __bf16* __restrict__ x,
                                     float* __restrict__ sums,  // [C][2]
                                     const float* __restrict__ mean_rstd,
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ beta, int C,
                                     int64_t spatial,
                                     const __bf16* __restrict__ res = nullptr) {
  const int nc = blockIdx.y;
  const int c = nc % C;
  const float mean = mean_rstd[c * 2], rstd = mean_rstd[c * 2 + 1];
  const float g = gamma[c], bt = beta[c];
  const __bf16* dyp = dy + (int64_t)nc * spatial;
  const __bf16* xp = x + (int64_t)nc * spatial;
  const __bf16* rp = RES ? res + (int64_t)nc * spatial : nullptr;
  float s = 0.f, sx = 0.f;
  int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (; i + 8 <= spatial; i += stride) {
    bf16x8v dv = *reinterpret_cast<const bf16x8v*>(dyp + i);
    bf16x8v xv = *reinterpret_cast<const bf16x8v*>(xp + i);
    bf16x8v rv;
    if (RES) rv = *reinterpret_cast<const bf16x8v*>(rp + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xh = ((float)xv[j] - mean) * rstd;
      float d = (float)dv[j];
      float z = g * xh + bt;
      if (RES) z += (float)rv[j];
      if (RELU && z <= 0.f) continue;
      s += d;
      sx += d * xh;
    }
  }
  if (i < spatial)
    for (int64_t j = i; j < spatial; ++j) {
      float xh = ((float)xp[j] - mean) * rstd;
      float d = (float)dyp[j];
      float z = g * xh + bt;
      if (RES) z += (float)rp[j];
      if (RELU && z <= 0.f) continue;
      s += d;
      sx += d * xh;
    }
  block_reduce_2(s, sx, &sums[c * 2], &sums[c * 2 + 1]);
}

// ---- bwd pass 2: dx = g*rstd*(dz - mean(dz) - xhat*mean(dz*xhat)) ---------
// RES: also writes dres = relu-masked dy (the residual branch gradient).
template <bool RELU, bool RES = false>
__global__ void bn_bwd_dx_kernel(const __bf16* __restrict__ dy,
                                 const __bf16* __restrict__ x,
                                 __bf16* __restrict__ dx,
                                 const float* __restrict__ mean_rstd,
                                 const float* __restrict__ sums,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 int64_t per_ch, int C, int64_t spatial,
                                 const __bf16* __restrict__ res = nullptr,
                                 __bf16* __restrict__ dres = nullptr) {
  const int nc = blockIdx.y;
  const int c = nc % C;
  const float mean = mean_rstd[c * 2], rstd = mean_rstd[c * 2 + 1];
  const float g = gamma[c], bt = beta[c];
  const float m_dy = sums[c * 2] / per_ch;
  const float m_dyxh = sums[c * 2 + 1] / per_ch;
  const float scale = g * rstd;
  const __bf16* dyp = dy + (int64_t)nc * spatial;
  const __bf16* xp = x + (int64_t)nc * spatial;
  const __bf16* rp = RES ? res + (int64_t)nc * spatial : nullptr;
  __bf16* drp = RES ? dres + (int64_t)nc * spatial : nullptr;
  __bf16* dxp = dx + (int64_t)nc * spatial;
  int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (; i + 8 <= spatial; i += stride) {
    bf16x8v dv = *reinterpret_cast<const bf16x8v*>(dyp + i);
    bf16x8v xv = *reinterpret_cast<const bf16x8v*>(xp + i);
    bf16x8v rv;
    if (RES) rv = *reinterpret_cast<const bf16x8v*>(rp + i);
    bf16x8v o, dr;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xh = ((float)xv[j] - mean) * rstd;
      float d = (float)dv[j];
      float z = g * xh + bt;
      if (RES) z += (float)rv[j];
      if (RELU && z <= 0.f) d = 0.f;
      if (RES) dr[j] = (__bf16)d;
      o[j] = (__bf16)(scale * (d - m_dy - xh * m_dyxh));
    }
    *reinterpret_cast<bf16x8v*>(dxp + i) = o;
    if (RES) *reinterpret_cast<bf16x8v*>(drp + i) = dr;
  }
  if (i < spatial)
    for (int64_t j = i; j < spatial; ++j) {
      float xh = ((float)xp[j] - mean) * rstd;
      float d = (float)dyp[j];
      float z = g * xh + bt;
      if (RES) z += (float)rp[j];
      if (RELU && z <= 0.f) d = 0.f;
      if (RES) drp[j] = (__bf16)d;
      dxp[j] = (__bf16)(scale * (d - m_dy - xh * m_dyxh));
    }
}

// ---- host -----------------------------------------------------------------
static void bn_grid(int64_t spatial, int nc, dim3& grid, dim3& blk) {
  int64_t xblocks = (spatial + 256 * 8 - 1) / (256 * 8);
  int64_t cap = std::max<int64_t>(1, 4096 / std::max(nc, 1));
  if (xblocks > cap) xblocks = cap;
  grid = dim3((unsigned)xblocks, (unsigned)nc);
  blk = dim3(256);
}

std::vector<torch::Tensor> bn3d_fwd(torch::Tensor x, torch::Tensor gamma,
                                    torch::Tensor beta, double eps,
                                    bool relu) {
  CHECK_GPU(x);
  auto xc = x.contiguous();
  TORCH_CHECK(xc.scalar_type() == torch::kBFloat16, "bn3d_fwd wants bf16");
  int N = (int)xc.size(0), C = (int)xc.size(1);
  int64_t spatial = xc.numel() / ((int64_t)N * C);
  auto g = gamma.to(torch::kFloat32).contiguous();
  auto b = beta.to(torch::kFloat32).contiguous();

  auto sums = torch::zeros({C, 2}, xc.options().dtype(torch::kFloat32));
  dim3 grid, blk;
  bn_grid(spatial, N * C, grid, blk);
  hipLaunchKernelGGL(bn_reduce_kernel, grid, blk, 0, current_stream(),
                     reinterpret_cast<const __bf16*>(xc.data_ptr()),
                     sums.data_ptr<float>(), C, spatial);
  int64_t per_ch = (int64_t)N * spatial;
  auto mean = sums.select(1, 0) / (double)per_ch;
  auto var = sums.select(1, 1) / (double)per_ch - mean * mean;
  auto rstd = torch::rsqrt(var.clamp_min(0) + eps);
  auto mean_rstd = torch::stack({mean, rstd}, 1).contiguous();

  auto y = torch::empty_like(xc);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, blk, 0, current_stream(),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<__bf16*>(y.data_ptr()),
                       mean_rstd.data_ptr<float>(), g.data_ptr<float>(),
                       b.data_ptr<float>(), C, spatial,
                       (const __bf16*)nullptr);
  };
  if (relu) launch(bn_normalize_kernel<true>);
  else launch(bn_normalize_kernel<false>);
  return {y, mean, var, mean_rstd};
}

// normalize-only (training form with externally computed batch stats —
// the conv-epilogue-stats path): y = [relu](affine(x)).
torch::Tensor bn3d_normalize(torch::Tensor x, torch::Tensor mean_rstd,
                             torch::Tensor gamma, torch::Tensor beta,
                             bool relu) {
  CHECK_GPU(x);
  auto xc = x.contiguous();
  int N = (int)xc.size(0), C = (int)xc.size(1);
  int64_t spatial = xc.numel() / ((int64_t)N * C);
  auto g = gamma.to(torch::kFloat32).contiguous();
  auto b = beta.to(torch::kFloat32).contiguous();
  auto mr = mean_rstd.to(torch::kFloat32).contiguous();
  auto y = torch::empty_like(xc);
  dim3 grid, blk;
  bn_grid(spatial, N * C, grid, blk);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, blk, 0, current_stream(),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<__bf16*>(y.data_ptr()),
                       mr.data_ptr<float>(), g.data_ptr<float>(),
                       b.data_ptr<float>(), C, spatial,
                       (const __bf16*)nullptr);
  };
  if (relu) launch(bn_normalize_kernel<true>);
  else launch(bn_normalize_kernel<false>);
  return y;
}

// fused residual form: y = relu(affine(x) + res); returns stats too.
std::vector<torch::Tensor> bn3d_fwd_res(torch::Tensor x, torch::Tensor res,
                                        torch::Tensor gamma,
                                        torch::Tensor beta, double eps) {
  CHECK_GPU(x);
  auto xc = x.contiguous();
  auto rc = res.to(torch::kBFloat16).contiguous();
  TORCH_CHECK(xc.scalar_type() == torch::kBFloat16, "bf16 only");
  TORCH_CHECK(rc.sizes() == xc.sizes(), "residual shape mismatch");
  int N = (int)xc.size(0), C = (int)xc.size(1);
  int64_t spatial = xc.numel() / ((int64_t)N * C);
  auto g = gamma.to(torch::kFloat32).contiguous();
  auto b = beta.to(torch::kFloat32).contiguous();
  auto sums = torch::zeros({C, 2}, xc.options().dtype(torch::kFloat32));
  dim3 grid, blk;
  bn_grid(spatial, N * C, grid, blk);
  hipLaunchKernelGGL(bn_reduce_kernel, grid, blk, 0, current_stream(),
                     reinterpret_cast<const __bf16*>(xc.data_ptr()),
                     sums.data_ptr<float>(), C, spatial);
  int64_t per_ch = (int64_t)N * spatial;
  auto mean = sums.select(1, 0) / (double)per_ch;
  auto var = sums.select(1, 1) / (double)per_ch - mean * mean;
  auto rstd = torch::rsqrt(var.clamp_min(0) + eps);
  auto mean_rstd = torch::stack({mean, rstd}, 1).contiguous();
  auto y = torch::empty_like(xc);
  hipLaunchKernelGGL((bn_normalize_kernel<true, true>), grid, blk, 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(xc.data_ptr()),
                     reinterpret_cast<__bf16*>(y.data_ptr()),
                     mean_rstd.data_ptr<float>(), g.data_ptr<float>(),
                     b.data_ptr<float>(), C, spatial,
                     reinterpret_cast<const __bf16*>(rc.data_ptr()));
  return {y, mean, var, mean_rstd};
}

// stats only — for the fused conv+BN path (normalize-on-load): the
// normalize pass never runs, the next conv applies the affine on load.
std::vector<torch::Tensor> bn3d_stats(torch::Tensor x, double eps) {
  CHECK_GPU(x);
  auto xc = x.contiguous();
  TORCH_CHECK(xc.scalar_type() == torch::kBFloat16, "bn3d_stats wants bf16");
  int N = (int)xc.size(0), C = (int)xc.size(1);
  int64_t spatial = xc.numel() / ((int64_t)N * C);
  auto sums = torch::zeros({C, 2}, xc.options().dtype(torch::kFloat32));
  dim3 grid, blk;
  bn_grid(spatial, N * C, grid, blk);
  hipLaunchKernelGGL(bn_reduce_kernel, grid, blk, 0, current_stream(),
                     reinterpret_cast<const __bf16*>(xc.data_ptr()),
                     sums.data_ptr<float>(), C, spatial);
  int64_t per_ch = (int64_t)N * spatial;
  auto mean = sums.select(1, 0) / (double)per_ch;
  auto var = sums.select(1, 1) / (double)per_ch - mean * mean;
  auto rstd = torch::rsqrt(var.clamp_min(0) + eps);
  auto mean_rstd = torch::stack({mean, rstd}, 1).contiguous();
  return {mean, var, mean_rstd};
}

torch::Tensor bn3d_infer(torch::Tensor x, torch::Tensor gamma,
                         torch::Tensor beta, torch::Tensor running_mean,
                         torch::Tensor running_var, double eps, bool relu) {
  CHECK_GPU(x);
  auto xc = x.contiguous();
  int N = (int)xc.size(0), C = (int)xc.size(1);
  int64_t spatial = xc.numel() / ((int64_t)N * C);
  auto g = gamma.to(torch::kFloat32).contiguous();
  auto b = beta.to(torch::kFloat32).contiguous();
  auto mean = running_mean.to(torch::kFloat32);
  auto rstd = torch::rsqrt(running_var.to(torch::kFloat32) + eps);
  auto mean_rstd = torch::stack({mean, rstd}, 1).contiguous();
  auto y = torch::empty_like(xc);
  dim3 grid, blk;
  bn_grid(spatial, N * C, grid, blk);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, blk, 0, current_stream(),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<__bf16*>(y.data_ptr()),
                       mean_rstd.data_ptr<float>(), g.data_ptr<float>(),
                       b.data_ptr<float>(), C, spatial,
                       (const __bf16*)nullptr);
  };
  if (relu) launch(bn_normalize_kernel<true>);
  else launch(bn_normalize_kernel<false>);
  return y;
}

std::vector<torch::Tensor> bn3d_bwd(torch::Tensor dy, torch::Tensor x,
                                    torch::Tensor mean_rstd,
                                    torch::Tensor gamma, torch::Tensor beta,
                                    bool relu) {
  CHECK_GPU(dy);
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  int N = (int)dyc.size(0), C = (int)dyc.size(1);
  int64_t spatial = dyc.numel() / ((int64_t)N * C);
  auto g = gamma.to(torch::kFloat32).contiguous();
  auto b = beta.to(torch::kFloat32).contiguous();

  auto sums = torch::zeros({C, 2}, dyc.options().dtype(torch::kFloat32));
  dim3 grid, blk;
  bn_grid(spatial, N * C, grid, blk);
  auto launch_r = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, blk, 0, current_stream(),
                       reinterpret_cast<const __bf16*>(dyc.data_ptr()),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       sums.data_ptr<float>(), mean_rstd.data_ptr<float>(),
                       g.data_ptr<float>(), b.data_ptr<float>(), C, spatial,
                       (const __bf16*)nullptr);
  };
  if (relu) launch_r(bn_bwd_reduce_kernel<true>);
  else launch_r(bn_bwd_reduce_kernel<false>);

  auto dx = torch::empty_like(dyc);
  int64_t per_ch = (int64_t)N * spatial;
  auto launch_d = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, blk, 0, current_stream(),
                       reinterpret_cast<const __bf16*>(dyc.data_ptr()),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<__bf16*>(dx.data_ptr()),
                       mean_rstd.data_ptr<float>(), sums.data_ptr<float>(),
                       g.data_ptr<float>(), b.data_ptr<float>(), per_ch, C,
                       spatial, (const __bf16*)nullptr, (__bf16*)nullptr);
  };
  if (relu) launch_d(bn_bwd_dx_kernel<true>);
  else launch_d(bn_bwd_dx_kernel<false>);

  auto dbeta = sums.select(1, 0).clone();
  auto dgamma = sums.select(1, 1).clone();
  return {dx, dgamma, dbeta};
}

// backward with the reduction precomputed (conv-epilogue bnbwd path):
// runs ONLY the dx pass; sums2 = [C][2] (sum dz*mask, sum dz*xhat*mask).
std::vector<torch::Tensor> bn3d_bwd_pre(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor mean_rstd,
                                        torch::Tensor gamma,
                                        torch::Tensor beta, bool relu,
                                        torch::Tensor sums2) {
  CHECK_GPU(dy);
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  int N = (int)dyc.size(0), C = (int)dyc.size(1);
  int64_t spatial = dyc.numel() / ((int64_t)N * C);
  auto g = gamma.to(torch::kFloat32).contiguous();
  auto b = beta.to(torch::kFloat32).contiguous();
  auto sums = sums2.to(torch::kFloat32).contiguous();
  TORCH_CHECK(sums.numel() == 2 * C, "sums2 must be [C,2]");
  dim3 grid, blk;
  bn_grid(spatial, N * C, grid, blk);
  auto dx = torch::empty_like(dyc);
  int64_t per_ch = (int64_t)N * spatial;
  auto launch_d = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, blk, 0, current_stream(),
                       reinterpret_cast<const __bf16*>(dyc.data_ptr()),
                       reinterpret_cast<const __bf16*>(xc.data_ptr()),
                       reinterpret_cast<__bf16*>(dx.data_ptr()),
                       mean_rstd.data_ptr<float>(), sums.data_ptr<float>(),
                       g.data_ptr<float>(), b.data_ptr<float>(), per_ch, C,
                       spatial, (const __bf16*)nullptr, (__bf16*)nullptr);
  };
  if (relu) launch_d(bn_bwd_dx_kernel<true>);
  else launch_d(bn_bwd_dx_kernel<false>);
  auto dbeta = sums.select(1, 0).clone();
  auto dgamma = sums.select(1, 1).clone();
  return {dx, dgamma, dbeta};
}

// backward of the fused residual form; extra output dres = masked dy.
std::vector<torch::Tensor> bn3d_bwd_res(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor res,
                                        torch::Tensor mean_rstd,
                                        torch::Tensor gamma,
                                        torch::Tensor beta) {
  CHECK_GPU(dy);
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  auto rc = res.contiguous();
  int N = (int)dyc.size(0), C = (int)dyc.size(1);
  int64_t spatial = dyc.numel() / ((int64_t)N * C);
  auto g = gamma.to(torch::kFloat32).contiguous();
  auto b = beta.to(torch::kFloat32).contiguous();
  auto sums = torch::zeros({C, 2}, dyc.options().dtype(torch::kFloat32));
  dim3 grid, blk;
  bn_grid(spatial, N * C, grid, blk);
  hipLaunchKernelGGL((bn_bwd_reduce_kernel<true, true>), grid, blk, 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(dyc.data_ptr()),
                     reinterpret_cast<const __bf16*>(xc.data_ptr()),
                     sums.data_ptr<float>(), mean_rstd.data_ptr<float>(),
                     g.data_ptr<float>(), b.data_ptr<float>(), C, spatial,
                     reinterpret_cast<const __bf16*>(rc.data_ptr()));
  auto dx = torch::empty_like(dyc);
  auto dres = torch::empty_like(dyc);
  int64_t per_ch = (int64_t)N * spatial;
  hipLaunchKernelGGL((bn_bwd_dx_kernel<true, true>), grid, blk, 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(dyc.data_ptr()),
                     reinterpret_cast<const __bf16*>(xc.data_ptr()),
                     reinterpret_cast<__bf16*>(dx.data_ptr()),
                     mean_rstd.data_ptr<float>(), sums.data_ptr<float>(),
                     g.data_ptr<float>(), b.data_ptr<float>(), per_ch, C,
                     spatial,
                     reinterpret_cast<const __bf16*>(rc.data_ptr()),
                     reinterpret_cast<__bf16*>(dres.data_ptr()));
  auto dbeta = sums.select(1, 0).clone();
  auto dgamma = sums.select(1, 1).clone();
  return {dx, dgamma, dbeta, dres};
}
