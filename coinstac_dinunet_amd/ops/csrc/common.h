// Shared helpers for the gfx950 (CDNA4) kernels.
// Hardware model: 64-wide wavefronts, 256 CUs in 8 XCDs, HBM3E ~8 TB/s.
// Elementwise kernels: 256-thread blocks, 16B/lane vectorized, grid-stride
// with the grid capped so the launch fills the chip without oversubscribing
// (cdna_hip_programming.md Guideline 11).
#pragma once
#include <hip/hip_runtime.h>

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

// torch.cuda's current stream (torch-on-ROCm tracks streams under the
// masqueraded CUDA device type).
static inline hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

#define WAVE_SIZE 64
#define ELEM_BLOCK 256

static inline int elem_grid(int64_t n, int per_thread = 4) {
  int64_t blocks = (n + (int64_t)ELEM_BLOCK * per_thread - 1) /
                   ((int64_t)ELEM_BLOCK * per_thread);
  // 256 CUs x 8 blocks/CU; grid-stride covers the rest.
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

#define CHECK_GPU(x) TORCH_CHECK((x).is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")

#define HIP_OK(expr)                                            \
  do {                                                          \
    hipError_t _e = (expr);                                     \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)
