// MI355X-native swap-or-not shuffle (SURVEY §8f.1 — the committee-shuffle
// SHA256 consumer on the block-import path; algorithm restated from
// consensus/swap_or_not_shuffle/src/shuffle_list.rs, 90 rounds at mainnet,
// chain_spec.rs:632).
//
// GPU shape: rounds are sequential; WITHIN a round every swap decision is
// independent (it depends only on positions, and the touched pairs are
// disjoint). Per round: kernel 1 computes the pivot + all per-256-window
// source hashes (one lane each); kernel 2 runs one lane per candidate pair,
// reads its decision bit and swaps. ~(n/256 + n/2) lanes per round; the
// swap pass is HBM-bound, the hash pass VALU-bound.
#include "sha256.hh"
#include "m3x_ctx.hh"
#include "../../include/m3x_consensus.h"

using namespace m3x;

namespace {

// hash of seed(32) | round(1) | window(4, LE), plus lane 0 computes the
// round pivot from H(seed | round)
__global__ void k_shuffle_hashes(const uint8_t *__restrict__ seed,
                                 uint32_t round, uint64_t n_windows,
                                 uint64_t list_size,
                                 uint8_t *__restrict__ hashes,
                                 uint64_t *__restrict__ pivot_out) {
  uint64_t w = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint8_t buf[37];
  for (int i = 0; i < 32; i++) buf[i] = seed[i];
  buf[32] = (uint8_t)round;
  if (w == 0) {
    uint8_t digest[32];
    m3x::sha256_bytes(buf, 33, digest);
    uint64_t raw = 0;
    for (int b = 7; b >= 0; b--) raw = (raw << 8) | digest[b];
    *pivot_out = raw % list_size;
  }
  if (w >= n_windows) return;
  buf[33] = (uint8_t)(w & 0xff);
  buf[34] = (uint8_t)((w >> 8) & 0xff);
  buf[35] = (uint8_t)((w >> 16) & 0xff);
  buf[36] = (uint8_t)((w >> 24) & 0xff);
  m3x::sha256_bytes(buf, 37, hashes + 32 * w);
}

// one lane per candidate pair; pair layout follows the reference's two
// mirror loops exactly (disjoint pairs -> race-free parallel swaps)
__global__ void k_shuffle_swaps(uint32_t *__restrict__ input,
                                uint64_t list_size,
                                const uint8_t *__restrict__ hashes,
                                const uint64_t *__restrict__ pivot_ptr) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint64_t pivot = *pivot_ptr;
  uint64_t mirror1 = (pivot + 1) >> 1;
  uint64_t mirror2 = (pivot + list_size + 1) >> 1;
  uint64_t n2 = mirror2 > (pivot + 1) ? mirror2 - (pivot + 1) : 0;
  uint64_t i, j;
  if (t < mirror1) {
    i = t;
    j = pivot - t;
  } else if (t < mirror1 + n2) {
    uint64_t q = t - mirror1;
    i = pivot + 1 + q;
    j = (list_size - 1) - q;
  } else {
    return;
  }
  const uint8_t *src = hashes + 32 * (j >> 8);
  uint8_t byte_v = src[(j & 0xff) >> 3];
  if ((byte_v >> (j & 0x07)) & 1) {
    uint32_t a = input[i];
    input[i] = input[j];
    input[j] = a;
  }
}

} // namespace

extern "C" {

int32_t m3x_shuffle_list_dev(m3x_ctx *ctx, void *indices_dev,
                             uint64_t list_size, uint32_t rounds,
                             const uint8_t seed[32], int32_t forwards) {
  if (!ctx || list_size == 0 || list_size > (1ull << 24) || rounds == 0 ||
      rounds > 255)
    return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  uint64_t n_windows = ((list_size - 1) >> 8) + 1;
  uint64_t need = 32 + 32 * n_windows + 64;
  int rc = m3x::ensure_scratch(ctx, &ctx->scratch_b, &ctx->scratch_b_bytes,
                               need);
  if (rc != M3X_OK) return rc;
  uint8_t *seed_d = ctx->scratch_b;
  uint8_t *hashes_d = ctx->scratch_b + 32;
  uint64_t *pivot_d =
      reinterpret_cast<uint64_t *>(ctx->scratch_b + 32 + 32 * n_windows);
  M3X_HIP_CHECK(hipMemcpyAsync(seed_d, seed, 32, hipMemcpyHostToDevice,
                               ctx->stream));
  uint32_t hblocks = (uint32_t)((n_windows + 63) / 64);
  uint64_t max_pairs = list_size / 2 + 2;
  uint32_t sblocks = (uint32_t)((max_pairs + 255) / 256);
  int r = forwards ? 0 : (int)rounds - 1;
  for (;;) {
    hipLaunchKernelGGL(k_shuffle_hashes, dim3(hblocks), dim3(64), 0,
                       ctx->stream, seed_d, (uint32_t)r, n_windows, list_size,
                       hashes_d, pivot_d);
    hipLaunchKernelGGL(k_shuffle_swaps, dim3(sblocks), dim3(256), 0,
                       ctx->stream, (uint32_t *)indices_dev, list_size,
                       hashes_d, pivot_d);
    if (forwards) {
      if (++r == (int)rounds) break;
    } else {
      if (r == 0) break;
      r--;
    }
  }
  M3X_HIP_CHECK(hipStreamSynchronize(ctx->stream));
  return M3X_OK;
}

int32_t m3x_shuffle_list(m3x_ctx *ctx, uint32_t *indices, uint64_t list_size,
                         uint32_t rounds, const uint8_t seed[32],
                         int32_t forwards) {
  if (!ctx || list_size == 0 || list_size > (1ull << 24) || rounds == 0)
    return M3X_ERR_ARG;
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  void *dev;
  M3X_HIP_CHECK(hipMalloc(&dev, list_size * 4));
  if (hipMemcpy(dev, indices, list_size * 4, hipMemcpyHostToDevice) !=
      hipSuccess) {
    (void)hipFree(dev);
    return M3X_ERR_HIP;
  }
  int32_t rc = m3x_shuffle_list_dev(ctx, dev, list_size, rounds, seed,
                                    forwards);
  if (rc == M3X_OK &&
      hipMemcpy(indices, dev, list_size * 4, hipMemcpyDeviceToHost) !=
          hipSuccess)
    rc = M3X_ERR_HIP;
  (void)hipFree(dev);
  return rc;
}

} // extern "C"
