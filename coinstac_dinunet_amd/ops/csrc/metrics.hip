// K12/K13 — metric reduction kernels.
// Prf1a: the reference's y*2+pred case trick spawns 5 kernels + 4 D2H
// syncs per batch (metrics.py:158-170); here ONE pass with per-wave
// ballot popcounts and 4 atomics per block.
// ConfusionMatrix: atomic 2D histogram (LDS-staged when K*K fits).
#include "common.h"

__global__ void prf1a_kernel(const int64_t* __restrict__ pred,
                             const int64_t* __restrict__ true_,
                             int64_t* __restrict__ counts, int64_t n) {
  __shared__ int s_counts[4];
  if (threadIdx.x < 4) s_counts[threadIdx.x] = 0;
  __syncthreads();
  int tp = 0, fp = 0, tn = 0, fn = 0;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const bool p = pred[i] != 0, t = true_[i] != 0;
    tp += p && t;
    fp += p && !t;
    tn += !p && !t;
    fn += !p && t;
  }
  // wave reduce then LDS atomics, one global atomic per block per counter
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
    tp += __shfl_down(tp, off);
    fp += __shfl_down(fp, off);
    tn += __shfl_down(tn, off);
    fn += __shfl_down(fn, off);
  }
  if ((threadIdx.x % WAVE_SIZE) == 0) {
    atomicAdd(&s_counts[0], tp);
    atomicAdd(&s_counts[1], fp);
    atomicAdd(&s_counts[2], tn);
    atomicAdd(&s_counts[3], fn);
  }
  __syncthreads();
  if (threadIdx.x < 4)
    atomicAdd(reinterpret_cast<unsigned long long*>(&counts[threadIdx.x]),
              (unsigned long long)s_counts[threadIdx.x]);
}

__global__ void confusion_kernel(const int64_t* __restrict__ pred,
                                 const int64_t* __restrict__ true_,
                                 int64_t* __restrict__ mat, int64_t n, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  int* s_mat = reinterpret_cast<int*>(smem_raw);
  const int kk = K * K;
  for (int i = threadIdx.x; i < kk; i += blockDim.x) s_mat[i] = 0;
  __syncthreads();
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int t = (int)true_[i], p = (int)pred[i];
    if (t >= 0 && t < K && p >= 0 && p < K) atomicAdd(&s_mat[t * K + p], 1);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < kk; i += blockDim.x)
    if (s_mat[i])
      atomicAdd(reinterpret_cast<unsigned long long*>(&mat[i]),
                (unsigned long long)s_mat[i]);
}

torch::Tensor prf1a_counts(torch::Tensor pred, torch::Tensor true_) {
  CHECK_GPU(pred);
  auto p = pred.contiguous().to(torch::kInt64);
  auto t = true_.contiguous().to(torch::kInt64);
  int64_t n = p.numel();
  auto counts = torch::zeros({4}, p.options());
  hipLaunchKernelGGL(prf1a_kernel, dim3(elem_grid(n, 16)), dim3(ELEM_BLOCK),
                     0, current_stream(), p.data_ptr<int64_t>(),
                     t.data_ptr<int64_t>(), counts.data_ptr<int64_t>(), n);
  return counts;  // [tp, fp, tn, fn]
}

torch::Tensor confusion_matrix(torch::Tensor pred, torch::Tensor true_,
                               int64_t num_classes) {
  CHECK_GPU(pred);
  auto p = pred.contiguous().to(torch::kInt64);
  auto t = true_.contiguous().to(torch::kInt64);
  int64_t n = p.numel();
  int K = (int)num_classes;
  auto mat = torch::zeros({num_classes, num_classes}, p.options());
  size_t smem = (size_t)K * K * sizeof(int);
  TORCH_CHECK(smem <= 160 * 1024, "num_classes too large for LDS histogram");
  hipLaunchKernelGGL(confusion_kernel, dim3(elem_grid(n, 16)),
                     dim3(ELEM_BLOCK), smem, current_stream(),
                     p.data_ptr<int64_t>(), t.data_ptr<int64_t>(),
                     mat.data_ptr<int64_t>(), n, K);
  return mat;
}
