// MI355X-native SSZ merkleization (hot path #2) + context/buffer C-ABI.
//
// Replaces BeaconState::update_tree_hash_cache's SHA256 merkleize
// (consensus/types/src/beacon_state.rs:2031-2046; math restated from
// merkle_proof/src/lib.rs:9-14,68-100, mix_in_length from
// deposit_data_tree.rs:26-38, Validator layout from validator.rs:25-35,
// 2^40 registry limit from eth_spec.rs:404). Designed CDNA4-first: one
// validator / one two-to-one node per lane (wave64), the padding-block
// compression constant-folded, multi-level LDS reduction per workgroup,
// grids ≫256 blocks to fill 8 XCDs. Integer VALU workload — no MFMA, HBM
// traffic ≈96B per node (see DESIGN.md roofline).
#include "sha256.hh"
#include "m3x_ctx.hh"
#include "../../include/m3x_consensus.h"
#include <vector>

using namespace m3x;

// ------------------------------------------------------------------ kernels

// single-thread: Z[0]=0^32, Z[i]=H(Z[i-1]||Z[i-1]) (merkle_proof ZERO_NODES)
__global__ void k_zero_ladder(uint8_t *zeros /*65*32*/) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  uint32_t z[8];
#pragma unroll
  for (int i = 0; i < 8; i++) z[i] = 0;
  for (int j = 0; j < 32; j++) zeros[j] = 0;
  for (int lvl = 1; lvl <= 64; lvl++) {
    uint32_t o[8];
    sha256_node(z, z, o);
#pragma unroll
    for (int i = 0; i < 8; i++) z[i] = o[i];
    uint8_t *dst = zeros + 32 * lvl;
    for (int i = 0; i < 8; i++) {
      dst[4 * i] = (uint8_t)(z[i] >> 24);
      dst[4 * i + 1] = (uint8_t)(z[i] >> 16);
      dst[4 * i + 2] = (uint8_t)(z[i] >> 8);
      dst[4 * i + 3] = (uint8_t)z[i];
    }
  }
}

__device__ __forceinline__ uint32_t be_load_u8x4(const uint8_t *p) {
  return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) |
         ((uint32_t)p[2] << 8) | p[3];
}

// Validator hash_tree_root leaves: one lane = one validator (121B SSZ ->
// 32B root, 8 two-to-one hashes: pubkey root + 7 field-tree nodes).
// The block cooperatively stages its 256*121 contiguous bytes through LDS
// with coalesced u32 loads, then each lane assembles its record.
__global__ __launch_bounds__(256) void k_validator_leaves(
    const uint8_t *__restrict__ ssz, uint64_t n, uint8_t *__restrict__ out) {
  __shared__ uint8_t lds[256 * 121 + 4];
  uint64_t base_rec = (uint64_t)blockIdx.x * 256;
  const uint32_t block_bytes = 256 * 121; // 30976, divisible by 4
  uint64_t gbase = base_rec * 121;
  uint64_t total = n * 121;
  // coalesced staging (u32 granularity; source is 4-byte aligned per block)
  for (uint32_t i = threadIdx.x; i < block_bytes / 4; i += 256) {
    uint64_t goff = gbase + 4ull * i;
    uint32_t v = 0;
    if (goff + 4 <= total) {
      v = *reinterpret_cast<const uint32_t *>(ssz + goff);
    } else if (goff < total) {
      // ragged tail of the whole array
      uint32_t b0 = ssz[goff];
      uint32_t b1 = (goff + 1 < total) ? ssz[goff + 1] : 0;
      uint32_t b2 = (goff + 2 < total) ? ssz[goff + 2] : 0;
      v = b0 | (b1 << 8) | (b2 << 16);
    }
    *reinterpret_cast<uint32_t *>(lds + 4 * i) = v;
  }
  __syncthreads();
  uint64_t rec = base_rec + threadIdx.x;
  if (rec >= n) return;
  const uint8_t *v = lds + 121 * threadIdx.x;
  // field chunks (big-endian words for the hasher); layout validator.rs:25-35
  uint32_t c[8][8];
#pragma unroll
  for (int i = 0; i < 8; i++)
#pragma unroll
    for (int j = 0; j < 8; j++) c[i][j] = 0;
  // c0 = H(pk[0:32] || pk[32:48] + 16 zero bytes)
  {
    uint32_t l[8], r[8];
#pragma unroll
    for (int i = 0; i < 8; i++) l[i] = be_load_u8x4(v + 4 * i);
#pragma unroll
    for (int i = 0; i < 4; i++) r[i] = be_load_u8x4(v + 32 + 4 * i);
#pragma unroll
    for (int i = 4; i < 8; i++) r[i] = 0;
    sha256_node(l, r, c[0]);
  }
#pragma unroll
  for (int i = 0; i < 8; i++) c[1][i] = be_load_u8x4(v + 48 + 4 * i); // wc
  c[2][0] = be_load_u8x4(v + 80); // effective_balance LE bytes as-is
  c[2][1] = be_load_u8x4(v + 84);
  c[3][0] = (uint32_t)v[88] << 24; // slashed bool in byte 0 of the chunk
  c[4][0] = be_load_u8x4(v + 89);
  c[4][1] = be_load_u8x4(v + 93);
  c[5][0] = be_load_u8x4(v + 97);
  c[5][1] = be_load_u8x4(v + 101);
  c[6][0] = be_load_u8x4(v + 105);
  c[6][1] = be_load_u8x4(v + 109);
  c[7][0] = be_load_u8x4(v + 113);
  c[7][1] = be_load_u8x4(v + 117);
  // 8-leaf tree: 4 + 2 + 1 hashes
  uint32_t h01[8], h23[8], h45[8], h67[8], ha[8], hb[8], root[8];
  sha256_node(c[0], c[1], h01);
  sha256_node(c[2], c[3], h23);
  sha256_node(c[4], c[5], h45);
  sha256_node(c[6], c[7], h67);
  sha256_node(h01, h23, ha);
  sha256_node(h45, h67, hb);
  sha256_node(ha, hb, root);
  uint8_t *dst = out + 32 * rec;
  uint32_t *dst32 = reinterpret_cast<uint32_t *>(dst);
#pragma unroll
  for (int i = 0; i < 8; i++) dst32[i] = __builtin_bswap32(root[i]);
}

// Multi-level reduction: each block folds 2^levels consecutive input nodes
// (levels <= 9) into one output node through LDS double buffers. Missing
// trailing nodes are the zero ladder Z[base_level] (padding with ladder
// values is exactly the right-sparse semantics of merkle_proof::create).
__global__ __launch_bounds__(256) void k_reduce(
    const uint8_t *__restrict__ in, uint64_t n_in, uint8_t *__restrict__ out,
    uint32_t levels, uint32_t base_level, const uint8_t *__restrict__ zeros) {
  __shared__ uint32_t A[512][8];
  __shared__ uint32_t B[256][8];
  const uint32_t span = 1u << levels; // <= 512
  uint64_t base = (uint64_t)blockIdx.x * span;
  uint32_t zw[8];
#pragma unroll
  for (int i = 0; i < 8; i++)
    zw[i] = be_load_u8x4(zeros + 32 * base_level + 4 * i);
  // load span nodes (each thread loads up to 2)
  for (uint32_t s = threadIdx.x; s < span; s += 256) {
    uint64_t idx = base + s;
    if (idx < n_in) {
      const uint32_t *src = reinterpret_cast<const uint32_t *>(in + 32 * idx);
#pragma unroll
      for (int i = 0; i < 8; i++) A[s][i] = __builtin_bswap32(src[i]);
    } else {
#pragma unroll
      for (int i = 0; i < 8; i++) A[s][i] = zw[i];
    }
  }
  __syncthreads();
  uint32_t cnt = span;
  for (uint32_t l = 0; l < levels; l++) {
    uint32_t half = cnt >> 1;
    if (l % 2 == 0) { // A -> B
      for (uint32_t t = threadIdx.x; t < half; t += 256)
        sha256_node(A[2 * t], A[2 * t + 1], B[t]);
    } else { // B -> A
      for (uint32_t t = threadIdx.x; t < half; t += 256)
        sha256_node(B[2 * t], B[2 * t + 1], A[t]);
    }
    cnt = half;
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    const uint32_t *res = (levels % 2 == 1) ? B[0] : A[0];
    uint32_t *dst = reinterpret_cast<uint32_t *>(out + 32 * blockIdx.x);
#pragma unroll
    for (int i = 0; i < 8; i++) dst[i] = __builtin_bswap32(res[i]);
  }
}

// single-thread finalize: carry node from `from_level` to `to_depth` against
// the zero ladder (node is always the LEFT child — right-sparse tree), then
// optionally mix_in_length.
__global__ void k_finalize(const uint8_t *__restrict__ node_in,
                           uint32_t from_level, uint32_t to_depth,
                           int64_t mix_len, const uint8_t *__restrict__ zeros,
                           uint8_t *__restrict__ out32) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  uint32_t node[8];
  const uint32_t *src = reinterpret_cast<const uint32_t *>(node_in);
#pragma unroll
  for (int i = 0; i < 8; i++) node[i] = __builtin_bswap32(src[i]);
  for (uint32_t l = from_level; l < to_depth; l++) {
    uint32_t z[8], o[8];
#pragma unroll
    for (int i = 0; i < 8; i++) z[i] = be_load_u8x4(zeros + 32 * l + 4 * i);
    sha256_node(node, z, o);
#pragma unroll
    for (int i = 0; i < 8; i++) node[i] = o[i];
  }
  if (mix_len >= 0) {
    // length chunk: LE64(len) || zeros  (deposit_data_tree.rs:26-38)
    uint64_t len = (uint64_t)mix_len;
    uint32_t lc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    lc[0] = __builtin_bswap32((uint32_t)(len & 0xffffffffu));
    lc[1] = __builtin_bswap32((uint32_t)(len >> 32));
    uint32_t o[8];
    sha256_node(node, lc, o);
#pragma unroll
    for (int i = 0; i < 8; i++) node[i] = o[i];
  }
  uint32_t *dst = reinterpret_cast<uint32_t *>(out32);
#pragma unroll
  for (int i = 0; i < 8; i++) dst[i] = __builtin_bswap32(node[i]);
}

// ---------------------------------------------------------------- host side

namespace m3x {
void time_begin_s(m3x_ctx *ctx, int k, hipStream_t st) {
  if (!ctx->timing) return;
  if (!ctx->ev_a[k]) {
    (void)hipEventCreate(&ctx->ev_a[k]);
    (void)hipEventCreate(&ctx->ev_b[k]);
  }
  (void)hipEventRecord(ctx->ev_a[k], st);
}

void time_end_s(m3x_ctx *ctx, int k, hipStream_t st) {
  if (!ctx->timing) return;
  (void)hipEventRecord(ctx->ev_b[k], st);
  (void)hipEventSynchronize(ctx->ev_b[k]);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, ctx->ev_a[k], ctx->ev_b[k]);
  ctx->kernel_ms[k] += ms;
  ctx->kernel_launches[k]++;
}

void time_begin(m3x_ctx *ctx, int k) { time_begin_s(ctx, k, ctx->stream); }
void time_end(m3x_ctx *ctx, int k) { time_end_s(ctx, k, ctx->stream); }

int ensure_scratch(m3x_ctx *ctx, uint8_t **buf, uint64_t *cur,
                   uint64_t bytes) {
  if (*cur >= bytes) return M3X_OK;
  if (*buf) (void)hipFree(*buf);
  *buf = nullptr;
  *cur = 0;
  if (hipMalloc(buf, bytes) != hipSuccess) return M3X_ERR_NOMEM;
  *cur = bytes;
  return M3X_OK;
}
} // namespace m3x

extern "C" {

int32_t m3x_abi_version(void) { return 1; }

int32_t m3x_timing_enable(m3x_ctx *ctx, int32_t on) {
  if (!ctx) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  ctx->timing = on != 0;
  for (int i = 0; i < M3X_K_COUNT; i++) {
    ctx->kernel_ms[i] = 0;
    ctx->kernel_launches[i] = 0;
  }
  return M3X_OK;
}

int32_t m3x_kernel_ms(m3x_ctx *ctx, int32_t kernel_id, double *ms,
                      uint64_t *launches) {
  if (!ctx || kernel_id < 0 || kernel_id >= M3X_K_COUNT) return M3X_ERR_ARG;
  *ms = ctx->kernel_ms[kernel_id];
  *launches = ctx->kernel_launches[kernel_id];
  return M3X_OK;
}

/* carry a 32B node from `from_level` to `to_depth` against the zero ladder,
 * then optionally mix_in_length — the multi-GPU cap-finishing primitive. */
int32_t m3x_finalize_root(m3x_ctx *ctx, const uint8_t node[32],
                          uint32_t from_level, uint32_t to_depth,
                          int64_t mix_len, uint8_t out_root[32]) {
  if (!ctx) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  uint8_t *tmp = ctx->small_pool;
  if (hipMemcpy(tmp, node, 32, hipMemcpyHostToDevice) != hipSuccess)
    return M3X_ERR_HIP;
  hipLaunchKernelGGL(k_finalize, dim3(1), dim3(64), 0, ctx->stream, tmp,
                     from_level, to_depth, mix_len, ctx->zeros_dev, tmp + 32);
  int32_t r2 = M3X_OK;
  if (hipMemcpyAsync(out_root, tmp + 32, 32, hipMemcpyDeviceToHost,
                     ctx->stream) != hipSuccess)
    r2 = M3X_ERR_HIP;
  if (hipStreamSynchronize(ctx->stream) != hipSuccess) r2 = M3X_ERR_HIP;
  return r2;
}

int32_t m3x_ctx_create(m3x_ctx **out, int32_t device) {
  auto *ctx = new m3x_ctx();
  ctx->device = device;
  if (hipSetDevice(device) != hipSuccess) {
    delete ctx;
    return M3X_ERR_HIP;
  }
  if (hipStreamCreate(&ctx->stream) != hipSuccess ||
      hipStreamCreate(&ctx->stream2) != hipSuccess ||
      hipStreamCreate(&ctx->stream3) != hipSuccess ||
      hipEventCreateWithFlags(&ctx->ev_s2, hipEventDisableTiming) !=
          hipSuccess) {
    delete ctx;
    return M3X_ERR_HIP;
  }
  for (int i = 0; i < 24; i++)
    if (hipEventCreateWithFlags(&ctx->ev_pipe[i], hipEventDisableTiming) !=
        hipSuccess) {
      delete ctx;
      return M3X_ERR_HIP;
    }
  if (hipMalloc(&ctx->zeros_dev, 65 * 32) != hipSuccess) {
    delete ctx;
    return M3X_ERR_NOMEM;
  }
  if (hipMalloc(&ctx->small_pool, m3x_ctx::SMALL_POOL_IN + 64) !=
      hipSuccess) {
    delete ctx;
    return M3X_ERR_NOMEM;
  }
  hipLaunchKernelGGL(k_zero_ladder, dim3(1), dim3(64), 0, ctx->stream,
                     ctx->zeros_dev);
  if (hipStreamSynchronize(ctx->stream) != hipSuccess) {
    delete ctx;
    return M3X_ERR_HIP;
  }
  *out = ctx;
  return M3X_OK;
}

void m3x_ctx_destroy(m3x_ctx *ctx) {
  if (!ctx) return;
  if (ctx->zeros_dev) (void)hipFree(ctx->zeros_dev);
  if (ctx->small_pool) (void)hipFree(ctx->small_pool);
  if (ctx->scratch_a) (void)hipFree(ctx->scratch_a);
  if (ctx->scratch_b) (void)hipFree(ctx->scratch_b);
  if (ctx->stream) (void)hipStreamDestroy(ctx->stream);
  if (ctx->stream2) (void)hipStreamDestroy(ctx->stream2);
  if (ctx->stream3) (void)hipStreamDestroy(ctx->stream3);
  if (ctx->ev_s2) (void)hipEventDestroy(ctx->ev_s2);
  for (int i = 0; i < 24; i++)
    if (ctx->ev_pipe[i]) (void)hipEventDestroy(ctx->ev_pipe[i]);
  delete ctx;
}

int32_t m3x_dev_alloc(m3x_ctx *ctx, uint64_t bytes, void **dev_ptr) {
  if (!ctx || !dev_ptr) return M3X_ERR_ARG;
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  M3X_HIP_CHECK(hipMalloc(dev_ptr, bytes));
  return M3X_OK;
}

int32_t m3x_dev_free(m3x_ctx *ctx, void *dev_ptr) {
  if (!ctx) return M3X_ERR_ARG;
  M3X_HIP_CHECK(hipFree(dev_ptr));
  return M3X_OK;
}

int32_t m3x_h2d(m3x_ctx *ctx, void *dst_dev, const void *src_host,
                uint64_t bytes) {
  if (!ctx) return M3X_ERR_ARG;
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  M3X_HIP_CHECK(hipMemcpyAsync(dst_dev, src_host, bytes, hipMemcpyHostToDevice,
                               ctx->stream));
  M3X_HIP_CHECK(hipStreamSynchronize(ctx->stream));
  return M3X_OK;
}

int32_t m3x_d2h(m3x_ctx *ctx, void *dst_host, const void *src_dev,
                uint64_t bytes) {
  if (!ctx) return M3X_ERR_ARG;
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  M3X_HIP_CHECK(hipMemcpyAsync(dst_host, src_dev, bytes, hipMemcpyDeviceToHost,
                               ctx->stream));
  M3X_HIP_CHECK(hipStreamSynchronize(ctx->stream));
  return M3X_OK;
}

// Reduce an array of nodes to a single subtree root at depth `depth`,
// writing the 32B root into out_dev (device). Returns applied levels.
static int32_t reduce_to_root(m3x_ctx *ctx, const uint8_t *nodes_dev,
                              uint64_t n, uint32_t depth, uint8_t *out_dev) {
  // alternate between scratch_b halves for stage outputs
  const uint8_t *cur = nodes_dev;
  uint64_t m = n;
  uint32_t level = 0;
  uint8_t *ping = nullptr, *pong = nullptr;
  uint64_t need = (n + 511) / 512 * 32 + 64;
  int rc = ensure_scratch(ctx, &ctx->scratch_b, &ctx->scratch_b_bytes,
                          2 * need);
  if (rc != M3X_OK) return rc;
  ping = ctx->scratch_b;
  pong = ctx->scratch_b + need;
  while (level < depth && m > 1) {
    uint32_t levels = depth - level < 9 ? depth - level : 9;
    uint64_t span = 1ull << levels;
    uint64_t n_out = (m + span - 1) / span;
    uint8_t *dst = (cur == ping) ? pong : ping;
    m3x::time_begin(ctx, M3X_K_REDUCE);
    hipLaunchKernelGGL(k_reduce, dim3((uint32_t)n_out), dim3(256), 0,
                       ctx->stream, cur, m, dst, levels, level,
                       ctx->zeros_dev);
    m3x::time_end(ctx, M3X_K_REDUCE);
    cur = dst;
    m = n_out;
    level += levels;
  }
  // finalize: carry to full depth (and no mix here)
  hipLaunchKernelGGL(k_finalize, dim3(1), dim3(64), 0, ctx->stream, cur, level,
                     depth, (int64_t)-1, ctx->zeros_dev, out_dev);
  return M3X_OK;
}

int32_t m3x_merkleize_chunks_dev(m3x_ctx *ctx, const void *chunks_dev,
                                 uint64_t n_chunks, uint32_t depth,
                                 int64_t mix_len, uint8_t out_root[32]) {
  if (!ctx) return M3X_ERR_ARG;
  if (n_chunks > (1ull << depth) && depth < 63) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  uint8_t *root_dev = ctx->small_pool + m3x_ctx::SMALL_POOL_IN;
  int32_t rc;
  if (n_chunks == 0) {
    // Z[depth] then optional mix
    hipLaunchKernelGGL(k_finalize, dim3(1), dim3(64), 0, ctx->stream,
                       ctx->zeros_dev + 32 * depth, depth, depth, mix_len,
                       ctx->zeros_dev, root_dev);
  } else {
    rc = reduce_to_root(ctx, (const uint8_t *)chunks_dev, n_chunks, depth,
                        root_dev + 32);
    // root_dev points into ctx->small_pool — nothing to free on error
    if (rc != M3X_OK) return rc;
    hipLaunchKernelGGL(k_finalize, dim3(1), dim3(64), 0, ctx->stream,
                       root_dev + 32, depth, depth, mix_len, ctx->zeros_dev,
                       root_dev);
  }
  int32_t r2 = M3X_OK;
  if (hipMemcpyAsync(out_root, root_dev, 32, hipMemcpyDeviceToHost,
                     ctx->stream) != hipSuccess)
    r2 = M3X_ERR_HIP;
  if (hipStreamSynchronize(ctx->stream) != hipSuccess) r2 = M3X_ERR_HIP;
  return r2;
}

int32_t m3x_merkleize_chunks(m3x_ctx *ctx, const uint8_t *chunks,
                             uint64_t n_chunks, uint32_t depth,
                             int64_t mix_len, uint8_t out_root[32]) {
  if (!ctx) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  void *dev = nullptr;
  bool pooled = false;
  uint64_t bytes = n_chunks * 32;
  if (bytes) {
    if (bytes <= m3x_ctx::SMALL_POOL_IN) {
      dev = ctx->small_pool;
      pooled = true;
    } else {
      M3X_HIP_CHECK(hipMalloc(&dev, bytes));
    }
    if (hipMemcpy(dev, chunks, bytes, hipMemcpyHostToDevice) != hipSuccess) {
      if (!pooled) (void)hipFree(dev);
      return M3X_ERR_HIP;
    }
  }
  int32_t rc =
      m3x_merkleize_chunks_dev(ctx, dev, n_chunks, depth, mix_len, out_root);
  if (dev && !pooled) (void)hipFree(dev);
  return rc;
}

int32_t m3x_validator_subtree_root_dev(m3x_ctx *ctx, const void *ssz_dev,
                                       uint64_t n, uint32_t depth,
                                       uint8_t out_root[32]) {
  if (!ctx || (n > (1ull << depth) && depth < 63)) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  // leaves
  uint64_t n_pad = n ? n : 1;
  int rc = ensure_scratch(ctx, &ctx->scratch_a, &ctx->scratch_a_bytes,
                          n_pad * 32);
  if (rc != M3X_OK) return rc;
  uint8_t *root_dev;
  M3X_HIP_CHECK(hipMalloc(&root_dev, 64));
  if (n > 0) {
    uint32_t blocks = (uint32_t)((n + 255) / 256);
    m3x::time_begin(ctx, M3X_K_LEAVES);
    hipLaunchKernelGGL(k_validator_leaves, dim3(blocks), dim3(256), 0,
                       ctx->stream, (const uint8_t *)ssz_dev, n,
                       ctx->scratch_a);
    m3x::time_end(ctx, M3X_K_LEAVES);
    int32_t r = reduce_to_root(ctx, ctx->scratch_a, n, depth, root_dev);
    if (r != M3X_OK) {
      (void)hipFree(root_dev);
      return r;
    }
  } else {
    hipLaunchKernelGGL(k_finalize, dim3(1), dim3(64), 0, ctx->stream,
                       ctx->zeros_dev + 32 * depth, depth, depth, (int64_t)-1,
                       ctx->zeros_dev, root_dev);
  }
  int32_t r2 = M3X_OK;
  if (hipMemcpyAsync(out_root, root_dev, 32, hipMemcpyDeviceToHost,
                     ctx->stream) != hipSuccess)
    r2 = M3X_ERR_HIP;
  if (hipStreamSynchronize(ctx->stream) != hipSuccess) r2 = M3X_ERR_HIP;
  (void)hipFree(root_dev);
  return r2;
}

int32_t m3x_merkleize_validators_dev(m3x_ctx *ctx, const void *ssz_dev,
                                     uint64_t n, uint8_t out_root[32]) {
  // List[Validator, 2^40]: depth-40 tree + mix_in_length(n)
  uint8_t sub[32];
  int32_t rc = m3x_validator_subtree_root_dev(ctx, ssz_dev, n, 40, sub);
  if (rc != M3X_OK) return rc;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  uint8_t *tmp;
  M3X_HIP_CHECK(hipMalloc(&tmp, 64));
  if (hipMemcpy(tmp, sub, 32, hipMemcpyHostToDevice) != hipSuccess) {
    (void)hipFree(tmp);
    return M3X_ERR_HIP;
  }
  hipLaunchKernelGGL(k_finalize, dim3(1), dim3(64), 0, ctx->stream, tmp, 40,
                     40, (int64_t)n, ctx->zeros_dev, tmp + 32);
  int32_t r2 = M3X_OK;
  if (hipMemcpyAsync(out_root, tmp + 32, 32, hipMemcpyDeviceToHost,
                     ctx->stream) != hipSuccess)
    r2 = M3X_ERR_HIP;
  if (hipStreamSynchronize(ctx->stream) != hipSuccess) r2 = M3X_ERR_HIP;
  (void)hipFree(tmp);
  return r2;
}

int32_t m3x_merkleize_validators(m3x_ctx *ctx, const uint8_t *ssz, uint64_t n,
                                 uint8_t out_root[32]) {
  if (!ctx) return M3X_ERR_ARG;
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  void *dev = nullptr;
  uint64_t bytes = n * 121;
  // pad allocation to a multiple of 4 for the staging loads
  if (n) {
    M3X_HIP_CHECK(hipMalloc(&dev, (bytes + 3) & ~3ull));
    if (hipMemcpy(dev, ssz, bytes, hipMemcpyHostToDevice) != hipSuccess) {
      (void)hipFree(dev);
      return M3X_ERR_HIP;
    }
  }
  int32_t rc = m3x_merkleize_validators_dev(ctx, dev, n, out_root);
  if (dev) (void)hipFree(dev);
  return rc;
}

} // extern "C"

// Batched small-container merkleize: one thread folds one element's chunk
// group (<=32 chunks, depth = ceil_log2(count)) — the BeaconState's many
// tiny containers (Fork, headers, Eth1Data votes, checkpoints,
// HistoricalSummary, payload header) in ONE launch.
__global__ void k_merkleize_batch(const uint8_t *__restrict__ chunks,
                                  const uint32_t *__restrict__ offs,
                                  uint64_t n, const uint8_t *__restrict__ zeros,
                                  uint8_t *__restrict__ out) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint32_t c0 = offs[i], c1 = offs[i + 1];
  uint32_t cnt = c1 - c0;
  uint32_t node[32][8];
  for (uint32_t j = 0; j < cnt && j < 32; j++) {
    const uint32_t *src =
        reinterpret_cast<const uint32_t *>(chunks + 32ull * (c0 + j));
    for (int q = 0; q < 8; q++) node[j][q] = __builtin_bswap32(src[q]);
  }
  uint32_t depth = 0;
  while ((1u << depth) < cnt) depth++;
  uint32_t m = cnt, level = 0;
  while (level < depth) {
    uint32_t nx = (m + 1) / 2;
    for (uint32_t q = 0; q < nx; q++) {
      uint32_t zw[8];
      const uint32_t *r;
      if (2 * q + 1 < m) {
        r = node[2 * q + 1];
      } else {
        for (int b = 0; b < 8; b++)
          zw[b] = be_load_u8x4(zeros + 32 * level + 4 * b);
        r = zw;
      }
      uint32_t o[8];
      sha256_node(node[2 * q], r, o);
      for (int b = 0; b < 8; b++) node[q][b] = o[b];
    }
    m = nx;
    level++;
  }
  uint32_t *dst = reinterpret_cast<uint32_t *>(out + 32 * i);
  for (int b = 0; b < 8; b++) dst[b] = __builtin_bswap32(node[0][b]);
}

extern "C" int32_t m3x_merkleize_batch(m3x_ctx *ctx, const uint8_t *chunks,
                                       const uint32_t *offsets,
                                       uint64_t n_elems,
                                       uint8_t *out_roots) {
  if (!ctx || n_elems == 0) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  uint64_t in_bytes = (uint64_t)offsets[n_elems] * 32;
  uint64_t off_bytes = (n_elems + 1) * 4;
  uint64_t out_bytes = n_elems * 32;
  uint64_t need = in_bytes + off_bytes + out_bytes + 256;
  uint8_t *base;
  bool pooled = need <= m3x_ctx::SMALL_POOL_IN;
  if (pooled) {
    base = ctx->small_pool;
  } else {
    M3X_HIP_CHECK(hipMalloc(&base, need));
  }
  uint8_t *in_d = base;
  uint8_t *off_d = base + in_bytes;
  uint8_t *out_d = base + in_bytes + ((off_bytes + 31) & ~31ull);
  int32_t rc = M3X_OK;
  do {
    if (hipMemcpyAsync(in_d, chunks, in_bytes, hipMemcpyHostToDevice,
                       ctx->stream) != hipSuccess ||
        hipMemcpyAsync(off_d, offsets, off_bytes, hipMemcpyHostToDevice,
                       ctx->stream) != hipSuccess) {
      rc = M3X_ERR_HIP;
      break;
    }
    uint32_t blocks = (uint32_t)((n_elems + 63) / 64);
    hipLaunchKernelGGL(k_merkleize_batch, dim3(blocks), dim3(64), 0,
                       ctx->stream, in_d, (const uint32_t *)off_d, n_elems,
                       ctx->zeros_dev, out_d);
    if (hipMemcpyAsync(out_roots, out_d, out_bytes, hipMemcpyDeviceToHost,
                       ctx->stream) != hipSuccess) {
      rc = M3X_ERR_HIP;
      break;
    }
    if (hipStreamSynchronize(ctx->stream) != hipSuccess) rc = M3X_ERR_HIP;
  } while (0);
  if (!pooled) (void)hipFree(base);
  return rc;
}

// ------------------- incremental registry merkleize --------
// The cache stores all tree levels contiguously in HBM (level l at
// offs[l], counts cap>>l). Updates rehash dirty leaves + root paths only
// (identical concurrent writes to a shared parent are benign).

struct m3x_registry_cache {
  uint32_t cap_log2 = 0;
  uint64_t capacity = 0;
  uint64_t n = 0; // current list length (mix_in_length)
  uint8_t *levels = nullptr;   // sum_{l=0..cap_log2} (cap>>l) * 32 bytes
  uint64_t *dirty_a = nullptr; // ping-pong dirty index lists
  uint64_t *dirty_b = nullptr;
  uint8_t *recs = nullptr;     // staging for update records
  uint64_t recs_cap = 0;
};

namespace {

__device__ __forceinline__ uint64_t lvl_off(uint32_t cap_log2, uint32_t l) {
  // sum of counts of levels < l, in nodes: cap*(2 - 2^-(l-1)) ... computed
  // iteratively (l <= 24)
  uint64_t off = 0, c = 1ull << cap_log2;
  for (uint32_t q = 0; q < l; q++) {
    off += c;
    c >>= 1;
  }
  return off;
}

// hash one level range into the next (full build)
__global__ void k_cache_level(const uint8_t *__restrict__ src, uint64_t n_out,
                              uint8_t *__restrict__ dst) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n_out) return;
  uint32_t l[8], r[8], o[8];
  const uint32_t *a = reinterpret_cast<const uint32_t *>(src + 64 * i);
#pragma unroll
  for (int q = 0; q < 8; q++) l[q] = __builtin_bswap32(a[q]);
#pragma unroll
  for (int q = 0; q < 8; q++) r[q] = __builtin_bswap32(a[8 + q]);
  m3x::sha256_node(l, r, o);
  uint32_t *d = reinterpret_cast<uint32_t *>(dst + 32 * i);
#pragma unroll
  for (int q = 0; q < 8; q++) d[q] = __builtin_bswap32(o[q]);
}

// recompute leaf roots for m dirty records and scatter into level 0
__global__ void k_cache_update_leaves(const uint8_t *__restrict__ recs,
                                      const uint64_t *__restrict__ idx,
                                      uint64_t m, uint8_t *__restrict__ level0) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= m) return;
  const uint8_t *v = recs + 121 * t;
  uint32_t c[8][8];
#pragma unroll
  for (int i = 0; i < 8; i++)
#pragma unroll
    for (int j = 0; j < 8; j++) c[i][j] = 0;
  {
    uint32_t l[8], r[8];
#pragma unroll
    for (int i = 0; i < 8; i++) l[i] = be_load_u8x4(v + 4 * i);
#pragma unroll
    for (int i = 0; i < 4; i++) r[i] = be_load_u8x4(v + 32 + 4 * i);
#pragma unroll
    for (int i = 4; i < 8; i++) r[i] = 0;
    m3x::sha256_node(l, r, c[0]);
  }
#pragma unroll
  for (int i = 0; i < 8; i++) c[1][i] = be_load_u8x4(v + 48 + 4 * i);
  c[2][0] = be_load_u8x4(v + 80);
  c[2][1] = be_load_u8x4(v + 84);
  c[3][0] = (uint32_t)v[88] << 24;
  c[4][0] = be_load_u8x4(v + 89);
  c[4][1] = be_load_u8x4(v + 93);
  c[5][0] = be_load_u8x4(v + 97);
  c[5][1] = be_load_u8x4(v + 101);
  c[6][0] = be_load_u8x4(v + 105);
  c[6][1] = be_load_u8x4(v + 109);
  c[7][0] = be_load_u8x4(v + 113);
  c[7][1] = be_load_u8x4(v + 117);
  uint32_t h01[8], h23[8], h45[8], h67[8], ha[8], hb[8], root[8];
  m3x::sha256_node(c[0], c[1], h01);
  m3x::sha256_node(c[2], c[3], h23);
  m3x::sha256_node(c[4], c[5], h45);
  m3x::sha256_node(c[6], c[7], h67);
  m3x::sha256_node(h01, h23, ha);
  m3x::sha256_node(h45, h67, hb);
  m3x::sha256_node(ha, hb, root);
  uint32_t *d = reinterpret_cast<uint32_t *>(level0 + 32 * idx[t]);
#pragma unroll
  for (int q = 0; q < 8; q++) d[q] = __builtin_bswap32(root[q]);
}

// one propagation step: for each dirty node at level l, recompute its
// parent at level l+1 and emit the parent index
__global__ void k_cache_propagate(uint8_t *__restrict__ levels,
                                  uint32_t cap_log2, uint32_t l,
                                  const uint64_t *__restrict__ dirty,
                                  uint64_t m,
                                  uint64_t *__restrict__ dirty_next) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= m) return;
  uint64_t child = dirty[t];
  uint64_t parent = child >> 1;
  const uint8_t *src =
      levels + 32 * (lvl_off(cap_log2, l) + (parent << 1));
  uint8_t *dst = levels + 32 * (lvl_off(cap_log2, l + 1) + parent);
  uint32_t a[8], b[8], o[8];
  const uint32_t *s32 = reinterpret_cast<const uint32_t *>(src);
#pragma unroll
  for (int q = 0; q < 8; q++) a[q] = __builtin_bswap32(s32[q]);
#pragma unroll
  for (int q = 0; q < 8; q++) b[q] = __builtin_bswap32(s32[8 + q]);
  m3x::sha256_node(a, b, o);
  uint32_t *d32 = reinterpret_cast<uint32_t *>(dst);
#pragma unroll
  for (int q = 0; q < 8; q++) d32[q] = __builtin_bswap32(o[q]);
  dirty_next[t] = parent;
}

} // namespace

extern "C" {

void m3x_registry_cache_destroy(m3x_registry_cache *c) {
  if (!c) return;
  if (c->levels) (void)hipFree(c->levels);
  if (c->dirty_a) (void)hipFree(c->dirty_a);
  if (c->dirty_b) (void)hipFree(c->dirty_b);
  if (c->recs) (void)hipFree(c->recs);
  delete c;
}

int32_t m3x_registry_cache_create(m3x_ctx *ctx, const uint8_t *ssz,
                                  uint64_t n, m3x_registry_cache **out) {
  if (!ctx || !out || n > (1ull << 24)) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  auto *c = new m3x_registry_cache();
  uint64_t cap = 256;
  uint32_t cl = 8;
  while (cap < n) {
    cap <<= 1;
    cl++;
  }
  c->cap_log2 = cl;
  c->capacity = cap;
  c->n = n;
  uint64_t total_nodes = 0, q = cap;
  for (uint32_t l = 0; l <= cl; l++) {
    total_nodes += q;
    q >>= 1;
  }
  if (hipMalloc(&c->levels, total_nodes * 32) != hipSuccess) {
    delete c;
    return M3X_ERR_NOMEM;
  }
  // level 0: leaf roots for [0,n) then zero-ladder Z[0] (= zero chunk) pad
  M3X_HIP_CHECK(hipMemsetAsync(c->levels, 0, cap * 32, ctx->stream));
  if (n) {
    void *ssz_d;
    M3X_HIP_CHECK(hipMalloc(&ssz_d, (n * 121 + 3) & ~3ull));
    if (hipMemcpyAsync(ssz_d, ssz, n * 121, hipMemcpyHostToDevice,
                       ctx->stream) != hipSuccess) {
      (void)hipFree(ssz_d);
      m3x_registry_cache_destroy(c);
      return M3X_ERR_HIP;
    }
    uint32_t blocks = (uint32_t)((n + 255) / 256);
    hipLaunchKernelGGL(k_validator_leaves, dim3(blocks), dim3(256), 0,
                       ctx->stream, (const uint8_t *)ssz_d, n, c->levels);
    (void)hipStreamSynchronize(ctx->stream);
    (void)hipFree(ssz_d);
  }
  // build all levels (full capacity, zero-padded leaves make the ladder)
  uint64_t off = 0, cnt = cap;
  for (uint32_t l = 0; l < cl; l++) {
    uint64_t n_out = cnt >> 1;
    uint8_t *src = c->levels + off * 32;
    uint8_t *dst = c->levels + (off + cnt) * 32;
    uint32_t blocks = (uint32_t)((n_out + 255) / 256);
    hipLaunchKernelGGL(k_cache_level, dim3(blocks), dim3(256), 0, ctx->stream,
                       src, n_out, dst);
    off += cnt;
    cnt = n_out;
  }
  // dirty lists + record staging (grown on demand)
  if (hipMalloc(&c->dirty_a, 65536 * 8) != hipSuccess ||
      hipMalloc(&c->dirty_b, 65536 * 8) != hipSuccess) {
    m3x_registry_cache_destroy(c);
    return M3X_ERR_NOMEM;
  }
  if (hipStreamSynchronize(ctx->stream) != hipSuccess) {
    m3x_registry_cache_destroy(c);
    return M3X_ERR_HIP;
  }
  *out = c;
  return M3X_OK;
}

int32_t m3x_registry_cache_root(m3x_ctx *ctx, m3x_registry_cache *c,
                                uint8_t out_root[32]) {
  if (!ctx || !c) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  uint64_t top = 0, q = c->capacity;
  for (uint32_t l = 0; l < c->cap_log2; l++) {
    top += q;
    q >>= 1;
  }
  uint8_t *root_dev = ctx->small_pool + m3x_ctx::SMALL_POOL_IN;
  hipLaunchKernelGGL(k_finalize, dim3(1), dim3(64), 0, ctx->stream,
                     c->levels + top * 32, c->cap_log2, 40, (int64_t)c->n,
                     ctx->zeros_dev, root_dev);
  M3X_HIP_CHECK(hipMemcpyAsync(out_root, root_dev, 32, hipMemcpyDeviceToHost,
                               ctx->stream));
  M3X_HIP_CHECK(hipStreamSynchronize(ctx->stream));
  return M3X_OK;
}

int32_t m3x_registry_cache_update(m3x_ctx *ctx, m3x_registry_cache *c,
                                  const uint64_t *indices,
                                  const uint8_t *recs, uint64_t m,
                                  uint8_t out_root[32]) {
  if (!ctx || !c) return M3X_ERR_ARG;
  if (m > 65536) return M3X_ERR_ARG; // dirty-list capacity this round
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  if (m) {
    for (uint64_t i = 0; i < m; i++) {
      if (indices[i] >= c->capacity) return M3X_ERR_ARG;
      if (indices[i] + 1 > c->n) c->n = indices[i] + 1;
    }
    uint64_t need = m * 121;
    if (c->recs_cap < need) {
      if (c->recs) (void)hipFree(c->recs);
      c->recs = nullptr;
      c->recs_cap = 0;
      if (hipMalloc(&c->recs, (need + 3) & ~3ull) != hipSuccess)
        return M3X_ERR_NOMEM;
      c->recs_cap = need;
    }
    M3X_HIP_CHECK(hipMemcpyAsync(c->recs, recs, m * 121,
                                 hipMemcpyHostToDevice, ctx->stream));
    M3X_HIP_CHECK(hipMemcpyAsync(c->dirty_a, indices, m * 8,
                                 hipMemcpyHostToDevice, ctx->stream));
    uint32_t blocks = (uint32_t)((m + 63) / 64);
    hipLaunchKernelGGL(k_cache_update_leaves, dim3(blocks), dim3(64), 0,
                       ctx->stream, c->recs, c->dirty_a, m, c->levels);
    uint64_t *cur = c->dirty_a, *nxt = c->dirty_b;
    for (uint32_t l = 0; l < c->cap_log2; l++) {
      hipLaunchKernelGGL(k_cache_propagate, dim3(blocks), dim3(64), 0,
                         ctx->stream, c->levels, c->cap_log2, l, cur, m, nxt);
      uint64_t *t = cur;
      cur = nxt;
      nxt = t;
    }
  }
  return m3x_registry_cache_root(ctx, c, out_root);
}

} // extern "C"
