// K4 — fused Adam / SGD steps.
// The reference runs torch.optim.Adam, i.e. ~10 elementwise kernels per
// parameter per step (basetrainer.py:48-54). Here: ONE grid-stride kernel
// over the whole flat arena (engine.FlatOptimizer) or one launch per
// tensor for the list form. Memory-bound: 4 fp32 streams in (p, g, m, v),
// 3 out — vectorized 4-wide, 256-thread blocks (HBM3E-roofline shaped).
#include "common.h"

#include <unordered_map>

struct AdamHyper {
  float lr, beta1, beta2, eps, weight_decay;
  float bias1, bias2;  // 1 - beta^t corrections
};

// float4-vectorized body; tail handled scalar.
__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            int64_t n, int64_t n4, AdamHyper h) {
  float4* p4 = reinterpret_cast<float4*>(p);
  const float4* g4 = reinterpret_cast<const float4*>(g);
  float4* m4 = reinterpret_cast<float4*>(m);
  float4* v4 = reinterpret_cast<float4*>(v);
  const float inv_b1 = 1.0f / h.bias1;
  const float inv_b2 = 1.0f / h.bias2;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (int64_t)gridDim.x * blockDim.x) {
    float4 pv = p4[i], gv = g4[i], mv = m4[i], vv = v4[i];
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float pp = (&pv.x)[k], gg = (&gv.x)[k];
      gg += h.weight_decay * pp;
      float mm = h.beta1 * (&mv.x)[k] + (1.0f - h.beta1) * gg;
      float vvk = h.beta2 * (&vv.x)[k] + (1.0f - h.beta2) * gg * gg;
      float mhat = mm * inv_b1;
      float vhat = vvk * inv_b2;
      pp -= h.lr * mhat / (sqrtf(vhat) + h.eps);
      (&pv.x)[k] = pp;
      (&mv.x)[k] = mm;
      (&vv.x)[k] = vvk;
    }
    p4[i] = pv;
    m4[i] = mv;
    v4[i] = vv;
  }
  // tail
  for (int64_t i = (n4 << 2) + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float pp = p[i], gg = g[i];
    gg += h.weight_decay * pp;
    float mm = h.beta1 * m[i] + (1.0f - h.beta1) * gg;
    float vv = h.beta2 * v[i] + (1.0f - h.beta2) * gg * gg;
    p[i] = pp - h.lr * (mm * (1.0f / h.bias1)) /
                    (sqrtf(vv * (1.0f / h.bias2)) + h.eps);
    m[i] = mm;
    v[i] = vv;
  }
}

__global__ void sgd_kernel(float* __restrict__ p, const float* __restrict__ g,
                           float* __restrict__ buf, int64_t n, float lr,
                           float momentum, float weight_decay) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float gg = g[i] + weight_decay * p[i];
    if (momentum != 0.0f) {
      float b = momentum * buf[i] + gg;
      buf[i] = b;
      gg = b;
    }
    p[i] -= lr * gg;
  }
}

// float4 path only when every stream is 16B-aligned (arena views can sit
// at arbitrary 4B offsets).
static inline int64_t vec4_count(int64_t n, const void* a, const void* b,
                                 const void* c, const void* d) {
  auto mis = [](const void* p) { return ((uintptr_t)p & 15) != 0; };
  if (mis(a) || mis(b) || mis(c) || mis(d)) return 0;
  return n >> 2;
}

static AdamHyper make_hyper(double lr, double beta1, double beta2, double eps,
                            double weight_decay, int64_t step) {
  AdamHyper h;
  h.lr = (float)lr;
  h.beta1 = (float)beta1;
  h.beta2 = (float)beta2;
  h.eps = (float)eps;
  h.weight_decay = (float)weight_decay;
  h.bias1 = 1.0f - powf((float)beta1, (float)step);
  h.bias2 = 1.0f - powf((float)beta2, (float)step);
  return h;
}


// Multi-tensor form: ONE launch for the whole parameter list. Blocks map
// to (tensor, chunk) pairs through a small device table (the per-tensor
// launch form cost ~60 launches/step on ResNet-18 — pure overhead).
__global__ void adam_mt_kernel(const int64_t* __restrict__ ptrs,
                               const int* __restrict__ bmap,
                               int64_t chunk, AdamHyper h) {
  const int t = bmap[blockIdx.x * 2];
  const int64_t off = (int64_t)bmap[blockIdx.x * 2 + 1] * chunk;
  float* p = reinterpret_cast<float*>(ptrs[t * 5 + 0]);
  const float* g = reinterpret_cast<const float*>(ptrs[t * 5 + 1]);
  float* m = reinterpret_cast<float*>(ptrs[t * 5 + 2]);
  float* v = reinterpret_cast<float*>(ptrs[t * 5 + 3]);
  const int64_t n = ptrs[t * 5 + 4];
  const int64_t end = min(off + chunk, n);
  const float inv_b1 = 1.0f / h.bias1;
  const float inv_b2 = 1.0f / h.bias2;
  for (int64_t i = off + threadIdx.x; i < end; i += blockDim.x) {
    float pp = p[i], gg = g[i];
    gg += h.weight_decay * pp;
    float mm = h.beta1 * m[i] + (1.0f - h.beta1) * gg;
    float vv = h.beta2 * v[i] + (1.0f - h.beta2) * gg * gg;
    p[i] = pp - h.lr * (mm * inv_b1) / (sqrtf(vv * inv_b2) + h.eps);
    m[i] = mm;
    v[i] = vv;
  }
}

void fused_adam_flat(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                     torch::Tensor v, double lr, double beta1, double beta2,
                     double eps, double weight_decay, int64_t step) {
  CHECK_GPU(p); CHECK_CONTIG(p); CHECK_CONTIG(g);
  int64_t n = p.numel();
  auto h = make_hyper(lr, beta1, beta2, eps, weight_decay, step);
  hipStream_t stream = current_stream();
  hipLaunchKernelGGL(adam_kernel, dim3(elem_grid(n)), dim3(ELEM_BLOCK), 0,
                     stream, p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), n,
                     vec4_count(n, p.data_ptr(), g.data_ptr(), m.data_ptr(),
                                v.data_ptr()), h);
}

void fused_adam(std::vector<torch::Tensor> params,
                std::vector<torch::Tensor> grads,
                std::vector<torch::Tensor> exp_avgs,
                std::vector<torch::Tensor> exp_avg_sqs, double lr,
                double beta1, double beta2, double eps, double weight_decay,
                int64_t step) {
  const int T = (int)params.size();
  if (T == 0) return;
  AdamHyper h;
  h.lr = (float)lr; h.beta1 = (float)beta1; h.beta2 = (float)beta2;
  h.eps = (float)eps; h.weight_decay = (float)weight_decay;
  h.bias1 = 1.0f - powf((float)beta1, (float)step);
  h.bias2 = 1.0f - powf((float)beta2, (float)step);

  const int64_t chunk = 16384;
  auto ptrs_cpu = torch::empty({T, 5}, torch::dtype(torch::kInt64));
  auto* pc = ptrs_cpu.data_ptr<int64_t>();
  std::vector<int> bmap_v;
  for (int t = 0; t < T; ++t) {
    TORCH_CHECK(params[t].scalar_type() == torch::kFloat32 &&
                grads[t].scalar_type() == torch::kFloat32,
                "fused_adam wants fp32 params/grads");
    pc[t * 5 + 0] = (int64_t)params[t].data_ptr();
    pc[t * 5 + 1] = (int64_t)grads[t].data_ptr();
    pc[t * 5 + 2] = (int64_t)exp_avgs[t].data_ptr();
    pc[t * 5 + 3] = (int64_t)exp_avg_sqs[t].data_ptr();
    const int64_t n = params[t].numel();
    pc[t * 5 + 4] = n;
    const int64_t nch = (n + chunk - 1) / chunk;
    for (int64_t c = 0; c < nch; ++c) {
      bmap_v.push_back(t);
      bmap_v.push_back((int)c);
    }
  }
  const unsigned G = (unsigned)(bmap_v.size() / 2);
  if (G == 0) return;  // all params empty
  // the (ptrs, bmap) tables depend only on tensor pointers and sizes —
  // stable across steps — so cache the device copies (the per-step H2D
  // upload was ~10% of an MLP step)
  size_t key = 1469598103934665603ull;
  auto mix = [&](int64_t v) {
    key ^= (size_t)v;
    key *= 1099511628211ull;
  };
  // hash ALL pointers (p, g, m, v) + sizes: a fresh optimizer on the
  // same params re-creates state tensors and must miss the cache
  for (int64_t i = 0; i < (int64_t)T * 5; ++i) mix(pc[i]);
  static std::unordered_map<size_t, std::pair<torch::Tensor, torch::Tensor>>
      s_cache;
  auto it = s_cache.find(key);
  if (it == s_cache.end()) {
    if (s_cache.size() > 64) s_cache.clear();
    auto dev = params[0].device();
    auto ptrs = ptrs_cpu.to(dev);
    auto bmap = torch::from_blob(bmap_v.data(), {(int64_t)bmap_v.size()},
                                 torch::dtype(torch::kInt32))
                    .clone()
                    .to(dev);
    it = s_cache.emplace(key, std::make_pair(ptrs, bmap)).first;
  }
  hipLaunchKernelGGL(adam_mt_kernel, dim3(G), dim3(ELEM_BLOCK), 0,
                     current_stream(), it->second.first.data_ptr<int64_t>(),
                     it->second.second.data_ptr<int>(), chunk, h);
}

void fused_sgd(std::vector<torch::Tensor> params,
               std::vector<torch::Tensor> grads,
               std::vector<torch::Tensor> bufs, double lr, double momentum,
               double weight_decay) {
  hipStream_t stream = current_stream();
  for (size_t i = 0; i < params.size(); ++i) {
    auto p = params[i];
    auto g = grads[i].to(torch::kFloat32).contiguous();
    int64_t n = p.numel();
    hipLaunchKernelGGL(sgd_kernel, dim3(elem_grid(n, 1)), dim3(ELEM_BLOCK), 0,
                       stream, p.data_ptr<float>(), g.data_ptr<float>(),
                       bufs[i].data_ptr<float>(), n, (float)lr,
                       (float)momentum, (float)weight_decay);
  }
}
