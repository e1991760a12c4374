// Shared context for the m3x_consensus library (internal).
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <mutex>

struct m3x_ctx {
  int device = 0;
  hipStream_t stream = nullptr;
  // device-resident zero-hash ladder Z[0..64] (computed on GPU at create)
  uint8_t *zeros_dev = nullptr;
  // growable scratch buffers for merkle reduction stages
  uint8_t *scratch_a = nullptr;
  uint64_t scratch_a_bytes = 0;
  uint8_t *scratch_b = nullptr;
  uint64_t scratch_b_bytes = 0;
  std::mutex mu; // one context serializes its own calls
};

#define M3X_HIP_CHECK(expr)                                                    \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) return M3X_ERR_HIP;                                  \
  } while (0)

namespace m3x {
// ensure scratch buffer has at least `bytes`; returns 0 or M3X_ERR_*
int ensure_scratch(m3x_ctx *ctx, uint8_t **buf, uint64_t *cur, uint64_t bytes);
} // namespace m3x
