// K5/K7 — flat bucket pack/unpack.
// Replaces the reference's per-param GPU->CPU numpy round trip
// (tensorutils.py:44-47, learner.py:25-26) with device-resident copies
// into/out of one contiguous comm buffer. The primary path (engine
// FlatGradBuffer) needs no pack at all; these kernels serve the
// compression engines (PowerSGD rank-1 group, rankDAD factors) and any
// list-of-tensors <-> flat conversion.
//
// One kernel launch for ALL tensors: a device-side descriptor table
// (src ptr, dst offset, numel) is uploaded once; blocks grid-stride over a
// global element index and binary-search their segment.
#include "common.h"

struct Seg {
  const float* src;
  float* dst;
  int64_t offset;  // element offset of this segment in the flat buffer
  int64_t numel;
};

__global__ void pack_kernel(const Seg* __restrict__ segs, int nseg,
                            float* __restrict__ flat, int64_t total) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    // binary search the segment containing i
    int lo = 0, hi = nseg - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (segs[mid].offset <= i) lo = mid; else hi = mid - 1;
    }
    flat[i] = segs[lo].src[i - segs[lo].offset];
  }
}

__global__ void unpack_kernel(const Seg* __restrict__ segs, int nseg,
                              const float* __restrict__ flat, int64_t total) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int lo = 0, hi = nseg - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (segs[mid].offset <= i) lo = mid; else hi = mid - 1;
    }
    segs[lo].dst[i - segs[lo].offset] = flat[i];
  }
}

static std::vector<Seg> build_segs(const std::vector<torch::Tensor>& ts,
                                   bool as_dst) {
  std::vector<Seg> segs;
  int64_t off = 0;
  for (auto& t : ts) {
    TORCH_CHECK(t.is_contiguous() && t.scalar_type() == torch::kFloat32,
                "pack/unpack expects contiguous fp32 tensors");
    Seg s;
    s.src = as_dst ? nullptr : t.data_ptr<float>();
    s.dst = as_dst ? t.data_ptr<float>() : nullptr;
    s.offset = off;
    s.numel = t.numel();
    off += s.numel;
    segs.push_back(s);
  }
  return segs;
}

static torch::Tensor upload_segs(const std::vector<Seg>& segs,
                                 const torch::Device& dev) {
  auto bytes = (int64_t)(segs.size() * sizeof(Seg));
  auto host = torch::from_blob((void*)segs.data(), {bytes},
                               torch::TensorOptions().dtype(torch::kUInt8));
  // blocking copy: `segs` lives on the caller's stack
  return host.to(dev);
}

void pack_tensors(std::vector<torch::Tensor> tensors, torch::Tensor flat) {
  CHECK_GPU(flat); CHECK_CONTIG(flat);
  auto segs = build_segs(tensors, /*as_dst=*/false);
  int64_t total = flat.numel();
  auto dseg = upload_segs(segs, flat.device());
  hipLaunchKernelGGL(pack_kernel, dim3(elem_grid(total)), dim3(ELEM_BLOCK), 0,
                     current_stream(),
                     reinterpret_cast<const Seg*>(dseg.data_ptr()),
                     (int)segs.size(), flat.data_ptr<float>(), total);
}

void unpack_tensors(torch::Tensor flat, std::vector<torch::Tensor> tensors) {
  CHECK_GPU(flat); CHECK_CONTIG(flat);
  auto segs = build_segs(tensors, /*as_dst=*/true);
  int64_t total = flat.numel();
  auto dseg = upload_segs(segs, flat.device());
  hipLaunchKernelGGL(unpack_kernel, dim3(elem_grid(total)), dim3(ELEM_BLOCK),
                     0, current_stream(),
                     reinterpret_cast<const Seg*>(dseg.data_ptr()),
                     (int)segs.size(), flat.data_ptr<float>(), total);
}
