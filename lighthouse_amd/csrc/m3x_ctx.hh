// Shared context for the m3x_consensus library (internal).
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <mutex>

// named kernel slots for per-kernel HIP-event timing (m3x_kernel_ms)
enum m3x_kernel_id {
  M3X_K_LEAVES = 0,
  M3X_K_REDUCE = 1,
  M3X_K_FINALIZE = 2,
  M3X_K_BLS_PREPARE = 3,
  M3X_K_BLS_H2C = 4,
  M3X_K_BLS_MILLER = 5,
  M3X_K_BLS_REDUCE = 6,
  M3X_K_BLS_FINISH = 7,
  M3X_K_BLS_AGG = 8,
  M3X_K_COUNT = 9
};

struct m3x_ctx {
  int device = 0;
  hipStream_t stream = nullptr;
  hipStream_t stream2 = nullptr; // overlap of independent kernels
  hipStream_t stream3 = nullptr; // third pipeline stage
  hipEvent_t ev_s2 = nullptr;
  hipEvent_t ev_pipe[24] = {};   // chunk-pipeline dependencies
  // device-resident zero-hash ladder Z[0..64] (computed on GPU at create)
  uint8_t *zeros_dev = nullptr;
  // growable scratch buffers for merkle reduction stages
  uint8_t *scratch_a = nullptr;
  uint64_t scratch_a_bytes = 0;
  uint8_t *scratch_b = nullptr;
  uint64_t scratch_b_bytes = 0;
  // persistent pool for small inputs + 32B root slots (avoids per-call
  // hipMalloc/hipFree on the many tiny container merkleizations)
  uint8_t *small_pool = nullptr; // layout: [256 KiB input][64B root]
  static constexpr uint64_t SMALL_POOL_IN = 256 * 1024;
  // per-kernel cumulative time (ms) + launch counts since last reset,
  // measured with hipEvents on `stream`
  hipEvent_t ev_a[M3X_K_COUNT] = {};
  hipEvent_t ev_b[M3X_K_COUNT] = {};
  double kernel_ms[M3X_K_COUNT] = {};
  uint64_t kernel_launches[M3X_K_COUNT] = {};
  bool timing = false;
  std::recursive_mutex mu; // one context serializes its own calls
};

namespace m3x {
// wrap a launch with events when ctx->timing (call after the launch)
void time_begin(m3x_ctx *ctx, int k);
void time_end(m3x_ctx *ctx, int k);
void time_begin_s(m3x_ctx *ctx, int k, hipStream_t st);
void time_end_s(m3x_ctx *ctx, int k, hipStream_t st);
} // namespace m3x

#define M3X_HIP_CHECK(expr)                                                    \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) return M3X_ERR_HIP;                                  \
  } while (0)

namespace m3x {
// ensure scratch buffer has at least `bytes`; returns 0 or M3X_ERR_*
int ensure_scratch(m3x_ctx *ctx, uint8_t **buf, uint64_t *cur, uint64_t bytes);
} // namespace m3x
