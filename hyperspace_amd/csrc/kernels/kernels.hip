// CDNA4 (gfx950 / MI355X) data-plane kernels for hyperspace_amd.
//
// Pure HIP, written for wave64 / 64-wide ballots / LDS-resident counters —
// no CUDA-compat paths.  These implement the reference's Spark data-plane
// operators:
//   K2  murmur3 bucket hash (Spark Murmur3_x86_32-compatible, seed 42)
//   K3  stable LSD radix sort (8/9-bit digits, ballot multi-split
//       ranking, constant-byte pass skipping / 9-bit pass planning)
//   K4  segmented sorted merge join (count + emit)
//   K7  sorted-set membership (lineage delete filter)
//   K8  segmented min/max + bloom filter build/probe
//   K10 z-order bit interleave
// plus filter-scan compaction (select_range), device exclusive scan and
// row gather.
//
// All kernels are memory-bound streaming ops: 256-thread blocks (4 waves),
// grid-stride loops sized ≫ 256 workgroups to fill 8 XCDs; occupancy-based
// latency hiding (no software pipelining needed per the CDNA guide's
// regime rules for streaming ops).

#include <hip/hip_runtime.h>

#include <cassert>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <type_traits>

#include "kernels.h"

#define HIP_CHECK(expr)                                              \
  do {                                                               \
    hipError_t _e = (expr);                                          \
    if (_e != hipSuccess) {                                          \
      fprintf(stderr, "HIP error %s at %s:%d\n",                     \
              hipGetErrorString(_e), __FILE__, __LINE__);            \
      abort();                                                       \
    }                                                                \
  } while (0)

namespace hsk {

constexpr int THREADS = 256;
constexpr int WAVES = THREADS / 64;

__host__ __device__ static inline int64_t cdiv(int64_t a, int64_t b) {
  return (a + b - 1) / b;
}

static inline int grid_for(int64_t n, int64_t per_block = THREADS) {
  int64_t blocks = cdiv(n, per_block);
  if (blocks > 4096) blocks = 4096;  // grid-stride beyond
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// ---------------------------------------------------------------------------
// Murmur3 (Spark Murmur3_x86_32, seed 42) — bit-exact with
// ops/cpu_ref.py::murmur3_* (tested against Spark golden values).
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint32_t rotl32(uint32_t x, int r) {
  return (x << r) | (x >> (32 - r));
}

__device__ __forceinline__ uint32_t mix_k1(uint32_t k1) {
  k1 *= 0xCC9E2D51u;
  k1 = rotl32(k1, 15);
  k1 *= 0x1B873593u;
  return k1;
}

__device__ __forceinline__ uint32_t mix_h1(uint32_t h1, uint32_t k1) {
  h1 ^= k1;
  h1 = rotl32(h1, 13);
  h1 = h1 * 5 + 0xE6546B64u;
  return h1;
}

__device__ __forceinline__ uint32_t fmix(uint32_t h1, uint32_t len) {
  h1 ^= len;
  h1 ^= h1 >> 16;
  h1 *= 0x85EBCA6Bu;
  h1 ^= h1 >> 13;
  h1 *= 0xC2B2AE35u;
  h1 ^= h1 >> 16;
  return h1;
}

__device__ __forceinline__ uint32_t murmur_i32(uint32_t v, uint32_t seed) {
  return fmix(mix_h1(seed, mix_k1(v)), 4);
}

__device__ __forceinline__ uint32_t murmur_i64(uint64_t v, uint32_t seed) {
  uint32_t low = (uint32_t)v;
  uint32_t high = (uint32_t)(v >> 32);
  uint32_t h1 = mix_h1(seed, mix_k1(low));
  h1 = mix_h1(h1, mix_k1(high));
  return fmix(h1, 8);
}

template <typename T, bool IS64>
__global__ void k_murmur(const T* __restrict__ vals,
                         const uint8_t* __restrict__ valid,
                         uint32_t* __restrict__ h,
                         int64_t n, bool first, uint32_t seed) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint32_t s = first ? seed : h[i];
    if (valid != nullptr && !valid[i]) {
      // Spark HashPartitioning: a null child leaves the running hash
      // unchanged (Murmur3Hash.eval skips null inputs)
      h[i] = s;
      continue;
    }
    if (IS64)
      h[i] = murmur_i64((uint64_t)vals[i], s);
    else
      h[i] = murmur_i32((uint32_t)vals[i], s);
  }
}

__global__ void k_pmod(const uint32_t* __restrict__ h,
                       int32_t* __restrict__ out, int64_t n, int32_t nb) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int32_t signed_h = (int32_t)h[i];
    int32_t m = signed_h % nb;
    out[i] = m < 0 ? m + nb : m;
  }
}

void murmur3_column(const void* vals, int kind, const uint8_t* valid,
                    uint32_t* h, int64_t n, bool first, uint32_t seed,
                    hipStream_t stream) {
  int g = grid_for(n);
  if (kind == 1) {
    hipLaunchKernelGGL((k_murmur<uint64_t, true>), dim3(g), dim3(THREADS), 0,
                       stream, (const uint64_t*)vals, valid, h, n, first,
                       seed);
  } else {
    hipLaunchKernelGGL((k_murmur<uint32_t, false>), dim3(g), dim3(THREADS), 0,
                       stream, (const uint32_t*)vals, valid, h, n, first,
                       seed);
  }
}

void pmod_buckets(const uint32_t* h, int32_t* out, int64_t n,
                  int32_t num_buckets, hipStream_t stream) {
  hipLaunchKernelGGL(k_pmod, dim3(grid_for(n)), dim3(THREADS), 0, stream, h,
                     out, n, num_buckets);
}

// ---------------------------------------------------------------------------
// Order-preserving u64 key normalization
// ---------------------------------------------------------------------------

__global__ void k_norm_i64(const int64_t* in, uint64_t* out, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (uint64_t)in[i] ^ 0x8000000000000000ull;
}

template <typename T>
__global__ void k_norm_small_int(const T* in, uint64_t* out, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (uint64_t)(int64_t)in[i] ^ 0x8000000000000000ull;
}

__global__ void k_norm_f64(const double* in, uint64_t* out, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t bits = __double_as_longlong(in[i]);
    uint64_t mask =
        bits < 0 ? 0xFFFFFFFFFFFFFFFFull : 0x8000000000000000ull;
    out[i] = (uint64_t)bits ^ mask;
  }
}

__global__ void k_norm_f32(const float* in, uint64_t* out, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    // match the CPU reference: widen to f64 first, then flip
    int64_t bits = __double_as_longlong((double)in[i]);
    uint64_t mask =
        bits < 0 ? 0xFFFFFFFFFFFFFFFFull : 0x8000000000000000ull;
    out[i] = (uint64_t)bits ^ mask;
  }
}

void normalize_key(const void* vals, int dtype, uint64_t* out, int64_t n,
                   hipStream_t stream) {
  int g = grid_for(n);
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL(k_norm_i64, dim3(g), dim3(THREADS), 0, stream,
                         (const int64_t*)vals, out, n);
      break;
    case 1:
      hipLaunchKernelGGL(k_norm_small_int<int32_t>, dim3(g), dim3(THREADS), 0,
                         stream, (const int32_t*)vals, out, n);
      break;
    case 2:
      hipLaunchKernelGGL(k_norm_f64, dim3(g), dim3(THREADS), 0, stream,
                         (const double*)vals, out, n);
      break;
    case 3:
      hipLaunchKernelGGL(k_norm_f32, dim3(g), dim3(THREADS), 0, stream,
                         (const float*)vals, out, n);
      break;
    case 4:
      hipLaunchKernelGGL(k_norm_small_int<int16_t>, dim3(g), dim3(THREADS), 0,
                         stream, (const int16_t*)vals, out, n);
      break;
    case 5:
      hipLaunchKernelGGL(k_norm_small_int<int8_t>, dim3(g), dim3(THREADS), 0,
                         stream, (const int8_t*)vals, out, n);
      break;
    default:
      hipLaunchKernelGGL(k_norm_small_int<uint8_t>, dim3(g), dim3(THREADS),
                         0, stream, (const uint8_t*)vals, out, n);
  }
}

// ---------------------------------------------------------------------------
// Stable LSD radix sort (8- or 9-bit digits, wave64 ballot multi-split)
// ---------------------------------------------------------------------------

constexpr int RS_RADIX = 256;      // 8-bit digits
constexpr int RS_V = 2;            // keys per thread per tile
constexpr int RS_TILE = THREADS * RS_V;  // 1024 keys per tile
constexpr int RS_MAX_BLOCKS = 2048;  // 256 CUs x 8 blocks/CU

static inline int rs_num_blocks(int64_t n) {
  int64_t tiles = cdiv(n, RS_TILE);
  return (int)(tiles < RS_MAX_BLOCKS ? (tiles < 1 ? 1 : tiles)
                                     : RS_MAX_BLOCKS);
}

int64_t radix_sort_hist_size(int64_t n) {
  (void)n;
  // sized for the widest digit (9-bit / 512-bin passes)
  return (int64_t)512 * RS_MAX_BLOCKS;
}

__global__ void k_xor_or_reduce(const uint64_t* __restrict__ keys, int64_t n,
                                uint64_t* __restrict__ out_mask) {
  __shared__ uint64_t lds[THREADS];
  uint64_t k0 = keys[0];
  uint64_t acc = 0;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    acc |= keys[i] ^ k0;
  lds[threadIdx.x] = acc;
  __syncthreads();
  for (int off = THREADS / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) lds[threadIdx.x] |= lds[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0)
    atomicOr((unsigned long long*)out_mask, (unsigned long long)lds[0]);
}

template <int RADIX>
__global__ void k_rs_hist(const uint64_t* __restrict__ keys, int64_t n,
                          int shift, uint32_t* __restrict__ hist, int nb,
                          int64_t tiles_per_block) {
  // CONTIGUOUS tile->block ranges: block-major order must equal element
  // order or cross-pass LSD stability breaks.
  __shared__ uint32_t lh[RADIX];
  for (int d = threadIdx.x; d < RADIX; d += blockDim.x) lh[d] = 0;
  __syncthreads();
  int64_t e0 = (int64_t)blockIdx.x * tiles_per_block * RS_TILE;
  int64_t e1 = e0 + tiles_per_block * RS_TILE;
  if (e1 > n) e1 = n;
  for (int64_t i = e0 + threadIdx.x; i < e1; i += THREADS) {
    int d = (int)((keys[i] >> shift) & (RADIX - 1));
    atomicAdd(&lh[d], 1u);
  }
  __syncthreads();
  for (int d = threadIdx.x; d < RADIX; d += blockDim.x)
    hist[(int64_t)d * nb + blockIdx.x] = lh[d];
}

// single-block exclusive scan over u32 (hist is 256*nb <= 262144 elements)
__global__ void k_scan_u32(uint32_t* __restrict__ data, int64_t n) {
  constexpr int T = 1024;
  __shared__ uint32_t sums[T];
  int t = threadIdx.x;
  int64_t chunk = cdiv(n, T);
  int64_t start = t * chunk;
  int64_t end = start + chunk < n ? start + chunk : n;
  uint32_t s = 0;
  for (int64_t i = start; i < end; i++) s += data[i];
  sums[t] = s;
  __syncthreads();
  for (int off = 1; off < T; off <<= 1) {
    uint32_t v = (t >= off) ? sums[t - off] : 0;
    __syncthreads();
    sums[t] += v;
    __syncthreads();
  }
  uint32_t base = (t == 0) ? 0 : sums[t - 1];
  for (int64_t i = start; i < end; i++) {
    uint32_t v = data[i];
    data[i] = base;
    base += v;
  }
}

template <typename P, int RADIX>
__global__ void k_rs_scatter(const uint64_t* __restrict__ keys,
                             const P* __restrict__ payload,
                             uint64_t* __restrict__ okeys,
                             P* __restrict__ opayload, int64_t n,
                             int shift, const uint32_t* __restrict__ hist,
                             int nb, int64_t tiles_per_block) {
  // LDS-staged multi-item scatter: each thread carries RS_V keys
  // (striped rounds, preserving element order), the 1024-key tile is
  // counting-sorted into LDS by digit, then written out with
  // threadIdx-contiguous reads — global stores become per-digit-run
  // coalesced bursts instead of RADIX-way scattered singles, and the
  // per-tile barrier cost amortizes over RS_V x the keys (the round-1
  // k_rs_scatter showed 32.6% issue-stall on the scattered stores).
  // RADIX 256 = byte passes; RADIX 512 = 9-bit passes (one fewer pass
  // for <=27 varying bits at ~1.5x the LDS per block).
  static_assert(RADIX % THREADS == 0, "digits split evenly over threads");
  constexpr int D = RADIX / THREADS;      // digits owned per thread
  constexpr int DBITS = 32 - __builtin_clz((unsigned)RADIX);  // sentinel
  __shared__ uint32_t cur[RADIX];
  // per (round, wave, digit) counts, scanned IN PLACE into prefixes.
  // Counter width: every value is bounded by the tile size (1024) so
  // u16 suffices; the 512-bin variant uses it to halve its LDS (30 KB
  // at u32 -> 5 blocks/CU), the proven 256-bin path keeps u32 (u16
  // measured ~noise-level slower there; profiles/r2_summary.md).
  using cnt_t =
      typename std::conditional<(RADIX > 256), uint16_t, uint32_t>::type;
  __shared__ cnt_t cnt[RS_V][WAVES][RADIX];
  __shared__ cnt_t tile_total[RADIX];
  __shared__ cnt_t digit_start[RADIX];  // excl scan of tile_total
  __shared__ uint32_t wsum[WAVES];
  __shared__ uint64_t stage[RS_TILE];  // keys, then payload (reused)
  for (int d = threadIdx.x; d < RADIX; d += blockDim.x)
    cur[d] = hist[(int64_t)d * nb + blockIdx.x];
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int64_t t0 = (int64_t)blockIdx.x * tiles_per_block;
  int64_t t1 = t0 + tiles_per_block;
  for (int64_t tile = t0; tile < t1 && tile * RS_TILE < n; tile++) {
    int64_t ebase = tile * (int64_t)RS_TILE;
    int tile_n = (int)((n - ebase) < RS_TILE ? (n - ebase) : RS_TILE);
    uint64_t key[RS_V];
    P pl[RS_V];
    int dig[RS_V];
    int rank[RS_V];
    for (int x = threadIdx.x;
         x < RS_V * WAVES * RADIX * (int)sizeof(cnt_t) / 4;
         x += blockDim.x)
      ((uint32_t*)cnt)[x] = 0;  // word-packed zeroing
    __syncthreads();
#pragma unroll
    for (int v = 0; v < RS_V; v++) {
      // striped round v covers elements [v*256, (v+1)*256): round-major
      // order equals element order, so cross-round ranks stay stable
      int64_t i = ebase + (int64_t)v * THREADS + threadIdx.x;
      bool valid = i < n;
      key[v] = valid ? keys[i] : 0;
      pl[v] = valid ? payload[i] : P(0);
      int d = valid ? (int)((key[v] >> shift) & (RADIX - 1)) : RADIX;
      dig[v] = d;
      // DBITS-bit ballot multi-split (sentinel included, never counted)
      unsigned long long eq = ~0ull;
#pragma unroll
      for (int b = 0; b < DBITS; b++) {
        unsigned long long m = __ballot((d >> b) & 1);
        eq &= ((d >> b) & 1) ? m : ~m;
      }
      rank[v] = __popcll(eq & ((1ull << lane) - 1ull));
      if (valid && rank[v] == 0)
        cnt[v][wave][d] = (cnt_t)__popcll(eq);
    }
    __syncthreads();
    // D consecutive digits per thread: in-place (round, wave) prefix +
    // wave-level shfl scan of the per-thread sums into digit_start
    // (thread-major consecutive ownership keeps digit order)
    {
      uint32_t tot[D];
      uint32_t tsum = 0;
#pragma unroll
      for (int q = 0; q < D; q++) {
        int dd = threadIdx.x * D + q;
        cnt_t p = 0;
#pragma unroll
        for (int v = 0; v < RS_V; v++)
#pragma unroll
          for (int w = 0; w < WAVES; w++) {
            cnt_t c = cnt[v][w][dd];
            cnt[v][w][dd] = p;
            p = (cnt_t)(p + c);
          }
        tile_total[dd] = p;
        tot[q] = p;
        tsum += p;
      }
      uint32_t v = tsum;  // inclusive scan over this wave's 64 threads
#pragma unroll
      for (int off = 1; off < 64; off <<= 1) {
        uint32_t t = __shfl_up(v, off);
        if (lane >= off) v += t;
      }
      if (lane == 63) wsum[wave] = v;
      __syncthreads();
      uint32_t wbase = 0;
#pragma unroll
      for (int w = 0; w < WAVES; w++)
        if (w < wave) wbase += wsum[w];
      uint32_t base = wbase + v - tsum;  // exclusive over threads
#pragma unroll
      for (int q = 0; q < D; q++) {
        int dd = threadIdx.x * D + q;
        digit_start[dd] = (cnt_t)base;
        base += tot[q];
      }
    }
    __syncthreads();
    // counting-sort the tile into LDS (digit-major, stable)
    uint32_t tile_pos[RS_V];
#pragma unroll
    for (int v = 0; v < RS_V; v++)
      if (dig[v] < RADIX) {
        tile_pos[v] = digit_start[dig[v]] + cnt[v][wave][dig[v]]
                      + rank[v];
        stage[tile_pos[v]] = key[v];
      }
    __syncthreads();
    // coalesced write-out: slot j holds the tile's j-th digit-ordered
    // key; destination = digit's global cursor + offset within run
    uint32_t dst[RS_V];
#pragma unroll
    for (int v = 0; v < RS_V; v++) {
      int j = v * THREADS + threadIdx.x;
      if (j < tile_n) {
        uint64_t k2 = stage[j];
        int d2 = (int)((k2 >> shift) & (RADIX - 1));
        dst[v] = cur[d2] + j - digit_start[d2];
        okeys[dst[v]] = k2;
      }
    }
    __syncthreads();
#pragma unroll
    for (int v = 0; v < RS_V; v++)
      if (dig[v] < RADIX) stage[tile_pos[v]] = (uint64_t)pl[v];
    __syncthreads();
#pragma unroll
    for (int v = 0; v < RS_V; v++) {
      int j = v * THREADS + threadIdx.x;
      if (j < tile_n) opayload[dst[v]] = (P)stage[j];
    }
    __syncthreads();
#pragma unroll
    for (int q = 0; q < D; q++) {
      int dd = threadIdx.x * D + q;
      cur[dd] += tile_total[dd];
    }
    // next tile's reads of cur happen after its own barriers
  }
}

template <typename P>
static void radix_sort_impl(uint64_t* keys, P* payload, uint64_t* tmp_keys,
                            P* tmp_payload, uint32_t* hist,
                            uint64_t* d_mask, int64_t n,
                            hipStream_t stream) {
  if (n <= 1) return;
  assert(n < (int64_t)UINT32_MAX);
  int nb = rs_num_blocks(n);
  int64_t tpb = cdiv(cdiv(n, THREADS), nb);
  int g = grid_for(n);

  // which bytes actually vary? (constant-byte pass skipping)
  HIP_CHECK(hipMemsetAsync(d_mask, 0, sizeof(uint64_t), stream));
  hipLaunchKernelGGL(k_xor_or_reduce, dim3(g), dim3(THREADS), 0, stream,
                     keys, n, d_mask);
  uint64_t mask = 0;
  HIP_CHECK(hipMemcpyAsync(&mask, d_mask, sizeof(uint64_t),
                           hipMemcpyDeviceToHost, stream));
  HIP_CHECK(hipStreamSynchronize(stream));

  uint64_t* ka = keys;
  uint64_t* kb = tmp_keys;
  P* pa = payload;
  P* pb = tmp_payload;
  // pass planning: byte-aligned digits with constant-byte skipping vs
  // 9-bit digits anchored at the lowest varying bit — take whichever
  // needs fewer passes (26-bit bench keys: 4 byte passes -> 3 nine-bit
  // passes).  Both cover every varying bit exactly once in LSB->MSB
  // order, so LSD stability is preserved.
  int shifts8[8], n8 = 0;
  for (int s = 0; s < 64; s += 8)
    if ((mask >> s) & 255ull) shifts8[n8++] = s;
  int shifts9[8], n9 = 0;
  if (mask) {
    for (int s = __builtin_ctzll(mask); s < 64; s += 9)
      if ((mask >> s) & 511ull) shifts9[n9++] = s;
  }
  // 9-bit passes saved one pass for 26-bit keys but measured ~10%
  // slower end-to-end on MI355X (the scatter is wave-cap bound, and the
  // wider digit costs an extra ballot + 2-digit scan per thread) —
  // byte passes stay the default; HS_RS_9BIT=1 re-enables the wide
  // planner for future tuning (profiles/r2_summary.md).
  static const bool want9 = [] {
    const char* e = getenv("HS_RS_9BIT");
    return e && e[0] == '1';
  }();
  bool use9 = n9 < n8 && want9;
  const int* shifts = use9 ? shifts9 : shifts8;
  int npass = use9 ? n9 : n8;
  int radix = use9 ? 512 : 256;
  for (int pi = 0; pi < npass; pi++) {
    int shift = shifts[pi];
    HIP_CHECK(hipMemsetAsync(hist, 0,
                             (size_t)radix * nb * sizeof(uint32_t),
                             stream));
    if (use9) {
      hipLaunchKernelGGL((k_rs_hist<512>), dim3(nb), dim3(THREADS), 0,
                         stream, ka, n, shift, hist, nb, tpb);
      hipLaunchKernelGGL(k_scan_u32, dim3(1), dim3(1024), 0, stream, hist,
                         (int64_t)512 * nb);
      hipLaunchKernelGGL((k_rs_scatter<P, 512>), dim3(nb), dim3(THREADS),
                         0, stream, ka, pa, kb, pb, n, shift, hist, nb,
                         tpb);
    } else {
      hipLaunchKernelGGL((k_rs_hist<256>), dim3(nb), dim3(THREADS), 0,
                         stream, ka, n, shift, hist, nb, tpb);
      hipLaunchKernelGGL(k_scan_u32, dim3(1), dim3(1024), 0, stream, hist,
                         (int64_t)256 * nb);
      hipLaunchKernelGGL((k_rs_scatter<P, 256>), dim3(nb), dim3(THREADS),
                         0, stream, ka, pa, kb, pb, n, shift, hist, nb,
                         tpb);
    }
    uint64_t* tk = ka; ka = kb; kb = tk;
    P* tp = pa; pa = pb; pb = tp;
  }
  if (ka != keys) {
    HIP_CHECK(hipMemcpyAsync(keys, ka, n * sizeof(uint64_t),
                             hipMemcpyDeviceToDevice, stream));
    HIP_CHECK(hipMemcpyAsync(payload, pa, n * sizeof(P),
                             hipMemcpyDeviceToDevice, stream));
  }
}

void radix_sort_pairs(uint64_t* keys, int64_t* payload, uint64_t* tmp_keys,
                      int64_t* tmp_payload, uint32_t* hist, uint64_t* d_mask,
                      int64_t n, hipStream_t stream) {
  radix_sort_impl<int64_t>(keys, payload, tmp_keys, tmp_payload, hist,
                           d_mask, n, stream);
}

void radix_sort_pairs32(uint64_t* keys, int32_t* payload,
                        uint64_t* tmp_keys, int32_t* tmp_payload,
                        uint32_t* hist, uint64_t* d_mask, int64_t n,
                        hipStream_t stream) {
  radix_sort_impl<int32_t>(keys, payload, tmp_keys, tmp_payload, hist,
                           d_mask, n, stream);
}

// ---------------------------------------------------------------------------
// Device-wide exclusive scan (i64), 3-kernel two-level
// ---------------------------------------------------------------------------

constexpr int SCAN_MAX_BLOCKS = 2048;

static inline int scan_num_blocks(int64_t n) {
  int64_t per = cdiv(n, SCAN_MAX_BLOCKS);
  if (per < THREADS) per = THREADS;
  int64_t blocks = cdiv(n, per);
  return (int)(blocks < 1 ? 1 : blocks);
}

// block-level exclusive scan of a contiguous range; emits block total
__global__ void k_scan_partial(const int64_t* __restrict__ in,
                               int64_t* __restrict__ out, int64_t n,
                               int64_t per_block,
                               int64_t* __restrict__ block_tot) {
  __shared__ int64_t sums[THREADS];
  int t = threadIdx.x;
  int64_t b0 = (int64_t)blockIdx.x * per_block;
  int64_t b1 = b0 + per_block < n ? b0 + per_block : n;
  int64_t chunk = cdiv(b1 - b0, (int64_t)THREADS);
  int64_t start = b0 + t * chunk;
  int64_t end = start + chunk < b1 ? start + chunk : b1;
  if (start > b1) start = b1;
  int64_t s = 0;
  for (int64_t i = start; i < end; i++) s += in[i];
  sums[t] = s;
  __syncthreads();
  for (int off = 1; off < THREADS; off <<= 1) {
    int64_t v = (t >= off) ? sums[t - off] : 0;
    __syncthreads();
    sums[t] += v;
    __syncthreads();
  }
  if (t == THREADS - 1) block_tot[blockIdx.x] = sums[t];
  int64_t base = (t == 0) ? 0 : sums[t - 1];
  for (int64_t i = start; i < end; i++) {
    int64_t v = in[i];
    out[i] = base;
    base += v;
  }
}

__global__ void k_scan_block_tots(int64_t* __restrict__ tots, int nb,
                                  int64_t* __restrict__ total) {
  // single block: exclusive scan of up to SCAN_MAX_BLOCKS totals
  __shared__ int64_t sums[1024];
  int t = threadIdx.x;
  int chunk = (nb + 1023) / 1024;
  int start = t * chunk;
  int end = start + chunk < nb ? start + chunk : nb;
  if (start > nb) start = nb;
  int64_t s = 0;
  for (int i = start; i < end; i++) s += tots[i];
  sums[t] = s;
  __syncthreads();
  for (int off = 1; off < 1024; off <<= 1) {
    int64_t v = (t >= off) ? sums[t - off] : 0;
    __syncthreads();
    sums[t] += v;
    __syncthreads();
  }
  if (t == 1023 && total != nullptr) total[0] = sums[t];
  int64_t base = (t == 0) ? 0 : sums[t - 1];
  for (int i = start; i < end; i++) {
    int64_t v = tots[i];
    tots[i] = base;
    base += v;
  }
}

__global__ void k_scan_add_base(int64_t* __restrict__ out, int64_t n,
                                int64_t per_block,
                                const int64_t* __restrict__ block_tot) {
  int64_t base = block_tot[blockIdx.x];
  if (base == 0) return;
  int64_t b0 = (int64_t)blockIdx.x * per_block;
  int64_t b1 = b0 + per_block < n ? b0 + per_block : n;
  for (int64_t i = b0 + threadIdx.x; i < b1; i += blockDim.x) out[i] += base;
}

void exclusive_scan_i64(const int64_t* in, int64_t* out, int64_t n,
                        int64_t* total, hipStream_t stream) {
  if (n == 0) {
    if (total) HIP_CHECK(hipMemsetAsync(total, 0, 8, stream));
    return;
  }
  int nb = scan_num_blocks(n);
  int64_t per_block = cdiv(n, nb);
  // temp block totals: stream-ordered allocation — pool-backed (no
  // steady-state allocation cost), freed on the same stream, and safe
  // under multi-stream use from one host thread (a shared static here
  // would race between concurrent streams)
  int64_t* tots = nullptr;
  HIP_CHECK(hipMallocAsync(&tots, SCAN_MAX_BLOCKS * sizeof(int64_t),
                           stream));
  hipLaunchKernelGGL(k_scan_partial, dim3(nb), dim3(THREADS), 0, stream, in,
                     out, n, per_block, tots);
  hipLaunchKernelGGL(k_scan_block_tots, dim3(1), dim3(1024), 0, stream, tots,
                     nb, total);
  hipLaunchKernelGGL(k_scan_add_base, dim3(nb), dim3(THREADS), 0, stream,
                     out, n, per_block, tots);
  HIP_CHECK(hipFreeAsync(tots, stream));
}

// ---------------------------------------------------------------------------
// Segmented sorted merge join (K4)
// ---------------------------------------------------------------------------

__device__ __forceinline__ int64_t lower_bound_u64(
    const uint64_t* __restrict__ a, int64_t lo, int64_t hi, uint64_t v) {
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (a[mid] < v)
      lo = mid + 1;
    else
      hi = mid;
  }
  return lo;
}

__device__ __forceinline__ int64_t upper_bound_u64(
    const uint64_t* __restrict__ a, int64_t lo, int64_t hi, uint64_t v) {
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (a[mid] <= v)
      lo = mid + 1;
    else
      hi = mid;
  }
  return lo;
}

// Two-phase tile merge join.  The old design wrote three per-left-row
// i64 arrays (counts/starts/segment) then scanned and re-read them —
// ~19 GB of HBM traffic for a 268M-row probe side.  Both phases here
// recompute the (L2-resident) narrowed searches instead and only per-
// TILE totals touch HBM: phase 1 block-reduces each 256-row tile's pair
// count (n/256 i64 writes), an exclusive scan over tiles sizes the
// output, and phase 2 redoes the searches, prefix-sums the counts in
// LDS and writes the pairs at tile_offset + intra-tile prefix — output
// ordered by left row, bit-identical to the one-phase result.
//
// find_range: tile-narrowed search shared by both phases.  Both sides
// are sorted within a segment, so a left tile maps into a contiguous
// right window; two cooperative searches bound it and per-row probes
// stay inside the cache-resident window.
// rows per thread: amortizes the serial tile-narrow searches over 4x
// bigger tiles; consecutive rows per thread keep output left-row order
#define MJ_RPT 8
#define MJ_TILE ((int64_t)THREADS * MJ_RPT)

__device__ __forceinline__ int64_t mj_seg_of(
    const int64_t* __restrict__ lseg, int64_t n_seg, int64_t row) {
  int64_t lo = 0, hi = n_seg;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (lseg[mid + 1] <= row)
      lo = mid + 1;
    else
      hi = mid;
  }
  return lo;
}

// threads 0/1 compute the two window bounds in parallel (each bound is
// a serial dependent-load chain; splitting halves the critical path)
__device__ __forceinline__ void mj_tile_narrow(
    const uint64_t* __restrict__ lkeys, const uint64_t* __restrict__ rkeys,
    const int64_t* __restrict__ lseg, const int64_t* __restrict__ rseg,
    int64_t n_left, int64_t n_seg, int64_t base, int64_t* tile_seg,
    int64_t* tile_lo, int64_t* tile_hi) {
  int64_t last = base + MJ_TILE - 1 < n_left - 1 ? base + MJ_TILE - 1
                                                 : n_left - 1;
  int64_t s = mj_seg_of(lseg, n_seg, base);
  if (threadIdx.x == 0) {
    if (last < lseg[s + 1]) {
      *tile_seg = s;
      *tile_lo = lower_bound_u64(rkeys, rseg[s], rseg[s + 1], lkeys[base]);
    } else {
      *tile_seg = -1;  // tile spans a segment boundary: per-row fallback
    }
  } else {
    if (last < lseg[s + 1])
      *tile_hi = upper_bound_u64(rkeys, rseg[s], rseg[s + 1], lkeys[last]);
  }
}

// LDS staging capacity for the narrowed right window (u64 keys): most
// tiles map into a tiny window (avg = n_right * MJ_TILE / n_left), so
// per-row probes hit LDS instead of chasing L2/HBM latency
#define MJ_LDS_WIN 2048

// fill this thread's MJ_RPT consecutive rows: one binary search for the
// first row, forward-monotone searches for the rest (keys ascend, so
// each row's range starts at or after the previous row's).
// ``rbase``/``r_base_off``: the searched array and the global offset of
// its element 0 — (lwin, tile_lo) when the window is LDS-staged, else
// (rkeys, 0); a_out entries are global right indices either way.
__device__ __forceinline__ int64_t mj_thread_rows(
    const uint64_t* __restrict__ lkeys, const uint64_t* __restrict__ rbase,
    int64_t r_base_off, const int64_t* __restrict__ lseg,
    const int64_t* __restrict__ rseg, int64_t n_left, int64_t n_seg,
    int64_t first, int64_t tile_seg, int64_t tile_lo, int64_t tile_hi,
    int64_t* a_out, int64_t* c_out) {
  int64_t total = 0;
  int64_t a_prev = tile_lo - r_base_off;
  int64_t r1 = tile_hi - r_base_off;
  for (int k = 0; k < MJ_RPT; k++) {
    int64_t i = first + k;
    if (i >= n_left || (tile_seg >= 0 && i >= lseg[tile_seg + 1])) {
      c_out[k] = 0;
      continue;
    }
    if (tile_seg < 0) {
      // boundary tile: resolve the segment per row; reset the monotone
      // cursor on the first row (tile_lo is unset) and on segment change
      int64_t s = mj_seg_of(lseg, n_seg, i);
      int64_t s_start = rseg[s];
      if (k == 0 || s_start > a_prev) a_prev = s_start;
      r1 = rseg[s + 1];
    }
    uint64_t key = lkeys[i];
    int64_t a = lower_bound_u64(rbase, a_prev, r1, key);
    int64_t b = upper_bound_u64(rbase, a, r1, key);
    a_out[k] = a + r_base_off;
    c_out[k] = b - a;
    total += b - a;
    a_prev = a;
  }
  return total;
}

// cooperative LDS stage of the narrowed window; returns true when the
// per-row searches should probe lwin (tile fits one segment and the
// window fits LDS)
__device__ __forceinline__ bool mj_stage_window(
    const uint64_t* __restrict__ rkeys, uint64_t* lwin, int64_t tile_seg,
    int64_t tile_lo, int64_t tile_hi) {
  bool use_lds = tile_seg >= 0 && tile_hi - tile_lo <= MJ_LDS_WIN;
  if (use_lds) {
    for (int64_t j = threadIdx.x; j < tile_hi - tile_lo; j += blockDim.x)
      lwin[j] = rkeys[tile_lo + j];
  }
  __syncthreads();
  return use_lds;
}

__global__ void k_mj_tile_count(const uint64_t* __restrict__ lkeys,
                                const uint64_t* __restrict__ rkeys,
                                const int64_t* __restrict__ lseg,
                                const int64_t* __restrict__ rseg,
                                int64_t n_left, int64_t n_seg,
                                int64_t* __restrict__ tile_counts) {
  __shared__ int64_t tile_seg, tile_lo, tile_hi;
  __shared__ int64_t red[THREADS];
  __shared__ uint64_t lwin[MJ_LDS_WIN];
  int64_t n_tiles = cdiv(n_left, MJ_TILE);
  for (int64_t tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
    int64_t base = tile * MJ_TILE;
    if (threadIdx.x < 2)
      mj_tile_narrow(lkeys, rkeys, lseg, rseg, n_left, n_seg, base,
                     &tile_seg, &tile_lo, &tile_hi);
    __syncthreads();
    bool use_lds = mj_stage_window(rkeys, lwin, tile_seg, tile_lo,
                                   tile_hi);
    int64_t a_arr[MJ_RPT], c_arr[MJ_RPT];
    red[threadIdx.x] = mj_thread_rows(
        lkeys, use_lds ? lwin : rkeys, use_lds ? tile_lo : 0, lseg, rseg,
        n_left, n_seg, base + (int64_t)threadIdx.x * MJ_RPT, tile_seg,
        tile_lo, tile_hi, a_arr, c_arr);
    __syncthreads();
    for (int off = blockDim.x / 2; off > 0; off >>= 1) {
      if ((int)threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
      __syncthreads();
    }
    if (threadIdx.x == 0) tile_counts[tile] = red[0];
    __syncthreads();
  }
}

__global__ void k_mj_tile_emit(const uint64_t* __restrict__ lkeys,
                               const uint64_t* __restrict__ rkeys,
                               const int64_t* __restrict__ lseg,
                               const int64_t* __restrict__ rseg,
                               int64_t n_left, int64_t n_seg,
                               const int64_t* __restrict__ tile_offsets,
                               int64_t* __restrict__ out_l,
                               int64_t* __restrict__ out_r) {
  __shared__ int64_t tile_seg, tile_lo, tile_hi;
  __shared__ int64_t pfx[THREADS];
  __shared__ uint64_t lwin[MJ_LDS_WIN];
  int64_t n_tiles = cdiv(n_left, MJ_TILE);
  for (int64_t tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
    int64_t base = tile * MJ_TILE;
    if (threadIdx.x < 2)
      mj_tile_narrow(lkeys, rkeys, lseg, rseg, n_left, n_seg, base,
                     &tile_seg, &tile_lo, &tile_hi);
    __syncthreads();
    bool use_lds = mj_stage_window(rkeys, lwin, tile_seg, tile_lo,
                                   tile_hi);
    int64_t a_arr[MJ_RPT], c_arr[MJ_RPT];
    int64_t my_total = mj_thread_rows(
        lkeys, use_lds ? lwin : rkeys, use_lds ? tile_lo : 0, lseg, rseg,
        n_left, n_seg, base + (int64_t)threadIdx.x * MJ_RPT, tile_seg,
        tile_lo, tile_hi, a_arr, c_arr);
    // exclusive intra-tile prefix over per-thread totals (Hillis-Steele)
    pfx[threadIdx.x] = my_total;
    __syncthreads();
    for (int off = 1; off < (int)blockDim.x; off <<= 1) {
      int64_t v = (int)threadIdx.x >= off ? pfx[threadIdx.x - off] : 0;
      __syncthreads();
      pfx[threadIdx.x] += v;
      __syncthreads();
    }
    int64_t off_out = tile_offsets[tile] + pfx[threadIdx.x] - my_total;
    for (int k = 0; k < MJ_RPT; k++) {
      int64_t i = base + (int64_t)threadIdx.x * MJ_RPT + k;
      for (int64_t j = 0; j < c_arr[k]; j++) {
        out_l[off_out] = i;
        out_r[off_out] = a_arr[k] + j;
        off_out++;
      }
    }
    __syncthreads();
  }
}

int64_t merge_join_tile_size() { return MJ_TILE; }

// K4b: segmented two-sorted-run merge permutation.  Each bucket b's
// rows lie in [seg[b], seg[b+1]) as [A-run | B-run] with the boundary
// at split[b] (split[b] == seg[b+1] for single-run buckets).  Each
// row's merged position is its local rank plus its rank in the
// opposite run; A-rows stably precede equal B-rows.  One launch
// replaces a per-bucket sort loop when a scan merges the two sorted
// files an incremental refresh / two-group build leaves per bucket.
__global__ void k_run_merge_perm(const uint64_t* __restrict__ keys,
                                 const int64_t* __restrict__ seg,
                                 const int64_t* __restrict__ split,
                                 int64_t n, int64_t n_seg,
                                 int64_t* __restrict__ perm) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t s = mj_seg_of(seg, n_seg, i);
    int64_t a0 = seg[s], b1 = seg[s + 1], sp = split[s];
    uint64_t k = keys[i];
    int64_t pos;
    if (i < sp) {  // A row: count B elements strictly below it
      pos = (i - a0) + (lower_bound_u64(keys, sp, b1, k) - sp);
    } else {  // B row: count A elements at or below it
      pos = (i - sp) + (upper_bound_u64(keys, a0, sp, k) - a0);
    }
    perm[a0 + pos] = i;
  }
}

void run_merge_perm(const uint64_t* keys, const int64_t* seg,
                    const int64_t* split, int64_t n, int64_t n_seg,
                    int64_t* perm, hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL(k_run_merge_perm, dim3(grid_for(n)), dim3(THREADS),
                     0, stream, keys, seg, split, n, n_seg, perm);
}

void merge_join_tile_count(const uint64_t* lkeys, const uint64_t* rkeys,
                           const int64_t* lseg, const int64_t* rseg,
                           int64_t n_left, int64_t n_seg,
                           int64_t* tile_counts, hipStream_t stream) {
  if (n_left == 0) return;
  hipLaunchKernelGGL(k_mj_tile_count, dim3(grid_for(n_left)), dim3(THREADS),
                     0, stream, lkeys, rkeys, lseg, rseg, n_left, n_seg,
                     tile_counts);
}

void merge_join_tile_emit(const uint64_t* lkeys, const uint64_t* rkeys,
                          const int64_t* lseg, const int64_t* rseg,
                          int64_t n_left, int64_t n_seg,
                          const int64_t* tile_offsets, int64_t* out_l,
                          int64_t* out_r, hipStream_t stream) {
  if (n_left == 0) return;
  hipLaunchKernelGGL(k_mj_tile_emit, dim3(grid_for(n_left)), dim3(THREADS),
                     0, stream, lkeys, rkeys, lseg, rseg, n_left, n_seg,
                     tile_offsets, out_l, out_r);
}

// ---------------------------------------------------------------------------
// Filter scan: stable compaction of rows in a u64 range (select_range)
// ---------------------------------------------------------------------------

int64_t select_num_blocks(int64_t n) { return rs_num_blocks(n); }

__device__ __forceinline__ bool range_pred(uint64_t k, uint64_t lo,
                                           uint64_t hi, bool lo_incl,
                                           bool hi_incl) {
  bool ge = lo_incl ? (k >= lo) : (k > lo);
  bool le = hi_incl ? (k <= hi) : (k < hi);
  return ge && le;
}

__global__ void k_sel_count(const uint64_t* __restrict__ keys, int64_t n,
                            uint64_t lo, uint64_t hi, bool lo_incl,
                            bool hi_incl, int64_t* __restrict__ bcounts,
                            int nb, int64_t tiles_per_block) {
  __shared__ int64_t lds[THREADS];
  int64_t acc = 0;
  int64_t t0 = (int64_t)blockIdx.x * tiles_per_block;
  int64_t t1 = t0 + tiles_per_block;
  for (int64_t tile = t0; tile < t1 && tile * THREADS < n; tile++) {
    int64_t i = tile * (int64_t)THREADS + threadIdx.x;
    if (i < n && range_pred(keys[i], lo, hi, lo_incl, hi_incl)) acc++;
  }
  lds[threadIdx.x] = acc;
  __syncthreads();
  for (int off = THREADS / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) lds[threadIdx.x] += lds[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) bcounts[blockIdx.x] = lds[0];
}

__global__ void k_sel_emit(const uint64_t* __restrict__ keys, int64_t n,
                           uint64_t lo, uint64_t hi, bool lo_incl,
                           bool hi_incl, const int64_t* __restrict__ bases,
                           int64_t* __restrict__ out_idx, int nb,
                           int64_t tiles_per_block) {
  __shared__ int64_t cur;
  __shared__ uint32_t wave_cnt[WAVES];
  __shared__ uint32_t wave_pref[WAVES];
  if (threadIdx.x == 0) cur = bases[blockIdx.x];
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int64_t t0 = (int64_t)blockIdx.x * tiles_per_block;
  int64_t t1 = t0 + tiles_per_block;
  for (int64_t tile = t0; tile < t1 && tile * THREADS < n; tile++) {
    int64_t i = tile * (int64_t)THREADS + threadIdx.x;
    bool pred =
        i < n && range_pred(keys[i], lo, hi, lo_incl, hi_incl);
    __syncthreads();
    unsigned long long m = __ballot(pred);
    int rank = __popcll(m & ((1ull << lane) - 1ull));
    if (lane == 0) wave_cnt[wave] = (uint32_t)__popcll(m);
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t p = 0;
      for (int w = 0; w < WAVES; w++) {
        wave_pref[w] = p;
        p += wave_cnt[w];
      }
      wave_cnt[0] = p;  // tile total (reuse slot after prefix consumed)
    }
    __syncthreads();
    if (pred) out_idx[cur + wave_pref[wave] + rank] = i;
    __syncthreads();
    if (threadIdx.x == 0) cur += wave_cnt[0];
  }
}

void select_range_count(const uint64_t* keys, int64_t n, uint64_t lo,
                        uint64_t hi, bool lo_incl, bool hi_incl,
                        int64_t* block_counts, int64_t* total,
                        hipStream_t stream) {
  if (n == 0) {
    if (total) HIP_CHECK(hipMemsetAsync(total, 0, 8, stream));
    return;
  }
  int nb = (int)select_num_blocks(n);
  int64_t tpb = cdiv(cdiv(n, THREADS), nb);
  hipLaunchKernelGGL(k_sel_count, dim3(nb), dim3(THREADS), 0, stream, keys,
                     n, lo, hi, lo_incl, hi_incl, block_counts, nb, tpb);
  hipLaunchKernelGGL(k_scan_block_tots, dim3(1), dim3(1024), 0, stream,
                     block_counts, (int)nb, total);
}

void select_range_emit(const uint64_t* keys, int64_t n, uint64_t lo,
                       uint64_t hi, bool lo_incl, bool hi_incl,
                       const int64_t* block_bases, int64_t* out_idx,
                       hipStream_t stream) {
  if (n == 0) return;
  int nb = (int)select_num_blocks(n);
  int64_t tpb = cdiv(cdiv(n, THREADS), nb);
  hipLaunchKernelGGL(k_sel_emit, dim3(nb), dim3(THREADS), 0, stream, keys, n,
                     lo, hi, lo_incl, hi_incl, block_bases, out_idx, nb, tpb);
}

// ---------------------------------------------------------------------------
// Sorted-set membership (K7 lineage filter)
// ---------------------------------------------------------------------------

__global__ void k_isin(const int64_t* __restrict__ vals, int64_t n,
                       const int64_t* __restrict__ set, int64_t m,
                       bool* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t v = vals[i];
    int64_t lo = 0, hi = m;
    while (lo < hi) {
      int64_t mid = (lo + hi) >> 1;
      if (set[mid] < v)
        lo = mid + 1;
      else
        hi = mid;
    }
    out[i] = (lo < m) && (set[lo] == v);
  }
}

void isin_sorted(const int64_t* vals, int64_t n, const int64_t* sorted_set,
                 int64_t m, bool* out, hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL(k_isin, dim3(grid_for(n)), dim3(THREADS), 0, stream,
                     vals, n, sorted_set, m, out);
}

// ---------------------------------------------------------------------------
// Segmented min/max (K8 MinMax sketch)
// ---------------------------------------------------------------------------

__global__ void k_seg_minmax(const int64_t* __restrict__ vals,
                             const int64_t* __restrict__ seg_off,
                             int64_t n_seg, int64_t* __restrict__ mins,
                             int64_t* __restrict__ maxs) {
  __shared__ int64_t lmin[THREADS];
  __shared__ int64_t lmax[THREADS];
  for (int64_t s = blockIdx.x; s < n_seg; s += gridDim.x) {
    int64_t a = seg_off[s], b = seg_off[s + 1];
    int64_t mn = INT64_MAX, mx = INT64_MIN;
    for (int64_t i = a + threadIdx.x; i < b; i += blockDim.x) {
      int64_t v = vals[i];
      mn = v < mn ? v : mn;
      mx = v > mx ? v : mx;
    }
    lmin[threadIdx.x] = mn;
    lmax[threadIdx.x] = mx;
    __syncthreads();
    for (int off = THREADS / 2; off > 0; off >>= 1) {
      if (threadIdx.x < off) {
        if (lmin[threadIdx.x + off] < lmin[threadIdx.x])
          lmin[threadIdx.x] = lmin[threadIdx.x + off];
        if (lmax[threadIdx.x + off] > lmax[threadIdx.x])
          lmax[threadIdx.x] = lmax[threadIdx.x + off];
      }
      __syncthreads();
    }
    if (threadIdx.x == 0) {
      mins[s] = (b > a) ? lmin[0] : 0;
      maxs[s] = (b > a) ? lmax[0] : 0;
    }
    __syncthreads();
  }
}

void segmented_minmax(const int64_t* vals, const int64_t* seg_off,
                      int64_t n_seg, int64_t* mins, int64_t* maxs,
                      hipStream_t stream) {
  if (n_seg == 0) return;
  int g = (int)(n_seg < 4096 ? n_seg : 4096);
  hipLaunchKernelGGL(k_seg_minmax, dim3(g), dim3(THREADS), 0, stream, vals,
                     seg_off, n_seg, mins, maxs);
}

// ---------------------------------------------------------------------------
// Bloom filter (K8), double hashing h1 + i*h2 over Murmur3 — matches
// cpu_ref._bloom_hashes
// ---------------------------------------------------------------------------

__device__ __forceinline__ void bloom_h12(int64_t v, uint32_t& h1,
                                          uint32_t& h2) {
  h1 = murmur_i64((uint64_t)v, 0);
  h2 = murmur_i64((uint64_t)v, h1);
}

__global__ void k_bloom_build(const int64_t* __restrict__ vals, int64_t n,
                              uint64_t* __restrict__ words, int64_t m_bits,
                              int k) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint32_t h1, h2;
    bloom_h12(vals[i], h1, h2);
    for (int j = 0; j < k; j++) {
      uint64_t combined =
          ((uint64_t)h1 + (uint64_t)j * (uint64_t)h2) & 0x7FFFFFFFFFFFFFFFull;
      uint64_t pos = combined % (uint64_t)m_bits;
      atomicOr((unsigned long long*)&words[pos >> 6],
               1ull << (pos & 63));
    }
  }
}

__global__ void k_bloom_probe(const int64_t* __restrict__ vals, int64_t n,
                              const uint64_t* __restrict__ words,
                              int64_t m_bits, int k, bool* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint32_t h1, h2;
    bloom_h12(vals[i], h1, h2);
    bool all = true;
    for (int j = 0; j < k && all; j++) {
      uint64_t combined =
          ((uint64_t)h1 + (uint64_t)j * (uint64_t)h2) & 0x7FFFFFFFFFFFFFFFull;
      uint64_t pos = combined % (uint64_t)m_bits;
      all = (words[pos >> 6] >> (pos & 63)) & 1;
    }
    out[i] = all;
  }
}

void bloom_build(const int64_t* vals, int64_t n, uint64_t* words,
                 int64_t m_bits, int k, hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL(k_bloom_build, dim3(grid_for(n)), dim3(THREADS), 0,
                     stream, vals, n, words, m_bits, k);
}

void bloom_probe(const int64_t* vals, int64_t n, const uint64_t* words,
                 int64_t m_bits, int k, bool* out, hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL(k_bloom_probe, dim3(grid_for(n)), dim3(THREADS), 0,
                     stream, vals, n, words, m_bits, k, out);
}

// K9 device sketch-predicate eval: probe every value against every
// file's bloom filter in one launch; out[f] = any value may be in f.
__global__ void k_bloom_probe_many(const int64_t* __restrict__ vals,
                                   int64_t n_vals,
                                   const uint64_t* __restrict__ words,
                                   int64_t words_per_filter,
                                   int64_t n_filters, int64_t m_bits,
                                   int k, bool* __restrict__ out) {
  int64_t total = n_filters * n_vals;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       t < total; t += stride) {
    int64_t f = t / n_vals;
    int64_t v = t - f * n_vals;
    const uint64_t* w = words + f * words_per_filter;
    uint32_t h1, h2;
    bloom_h12(vals[v], h1, h2);
    bool all = true;
    for (int j = 0; j < k && all; j++) {
      uint64_t combined = ((uint64_t)h1 + (uint64_t)j * (uint64_t)h2) &
                          0x7FFFFFFFFFFFFFFFull;
      uint64_t pos = combined % (uint64_t)m_bits;
      all = (w[pos >> 6] >> (pos & 63)) & 1;
    }
    if (all) out[f] = true;  // benign same-value race
  }
}

void bloom_probe_many(const int64_t* vals, int64_t n_vals,
                      const uint64_t* words, int64_t words_per_filter,
                      int64_t n_filters, int64_t m_bits, int k, bool* out,
                      hipStream_t stream) {
  if (n_vals == 0 || n_filters == 0) return;
  hipLaunchKernelGGL(k_bloom_probe_many,
                     dim3(grid_for(n_filters * n_vals)), dim3(THREADS), 0,
                     stream, vals, n_vals, words, words_per_filter,
                     n_filters, m_bits, k, out);
}

// ---------------------------------------------------------------------------
// Z-order bit interleave (K10)
// ---------------------------------------------------------------------------

struct ZCols {
  const uint64_t* ptr[8];
};

__global__ void k_zorder(ZCols cols, int n_cols, int bits_per_col, int64_t n,
                         uint64_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t z = 0;
    for (int b = 0; b < bits_per_col; b++) {
      for (int c = 0; c < n_cols; c++) {
        uint64_t bit = (cols.ptr[c][i] >> (63 - b)) & 1ull;
        int out_pos = 63 - (b * n_cols + c);
        z |= bit << out_pos;
      }
    }
    out[i] = z;
  }
}

void zorder_key(const uint64_t* const* cols, int n_cols, int bits_per_col,
                int64_t n, uint64_t* out, hipStream_t stream) {
  if (n == 0) return;
  ZCols z{};
  for (int c = 0; c < n_cols && c < 8; c++) z.ptr[c] = cols[c];
  hipLaunchKernelGGL(k_zorder, dim3(grid_for(n)), dim3(THREADS), 0, stream,
                     z, n_cols, bits_per_col, n, out);
}

// ---------------------------------------------------------------------------
// Row gather
// ---------------------------------------------------------------------------

template <typename T>
__global__ void k_gather(const T* __restrict__ in,
                         const int64_t* __restrict__ idx,
                         T* __restrict__ out, int64_t n) {
  // simple grid-stride gather: measured EQUAL to explicit 4-way ILP
  // variants (37 Grows/s / ~0.9 TB/s effective on 268M random rows —
  // the compiler pipelines the independent iterations; the
  // 4-consecutive-per-thread form was 17% SLOWER from broken idx/out
  // coalescing, profiles/r2_summary.md)
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = in[idx[i]];
}

void gather(const void* in, const int64_t* idx, void* out, int64_t n,
            int elem_size, hipStream_t stream) {
  if (n == 0) return;
  int g = grid_for(n);
  switch (elem_size) {
    case 8:
      hipLaunchKernelGGL(k_gather<uint64_t>, dim3(g), dim3(THREADS), 0,
                         stream, (const uint64_t*)in, idx, (uint64_t*)out, n);
      break;
    case 4:
      hipLaunchKernelGGL(k_gather<uint32_t>, dim3(g), dim3(THREADS), 0,
                         stream, (const uint32_t*)in, idx, (uint32_t*)out, n);
      break;
    case 2:
      hipLaunchKernelGGL(k_gather<uint16_t>, dim3(g), dim3(THREADS), 0,
                         stream, (const uint16_t*)in, idx, (uint16_t*)out, n);
      break;
    default:
      hipLaunchKernelGGL(k_gather<uint8_t>, dim3(g), dim3(THREADS), 0,
                         stream, (const uint8_t*)in, idx, (uint8_t*)out, n);
  }
}

// ---------------------------------------------------------------------------
// K1: parquet RLE/bit-packed hybrid decode (dictionary indices)
// ---------------------------------------------------------------------------

__global__ void k_rle_decode(const uint8_t* __restrict__ src,
                             const int64_t* __restrict__ run_kind,
                             const int64_t* __restrict__ run_out_off,
                             const int64_t* __restrict__ run_len,
                             const int64_t* __restrict__ run_bitoff,
                             const int64_t* __restrict__ run_value,
                             int64_t n_runs, int bit_width,
                             uint32_t* __restrict__ out, int64_t n_out) {
  // one WAVE per run: lanes cooperate on literal bit-unpacking and
  // repeated fills (runs are short — parquet literal groups are 8
  // values, writers emit up to ~512-value runs)
  int64_t waves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  int64_t wid = (((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6);
  int lane = threadIdx.x & 63;
  uint32_t mask = bit_width >= 32 ? 0xFFFFFFFFu
                                  : ((1u << bit_width) - 1u);
  for (int64_t r = wid; r < n_runs; r += waves) {
    int64_t off = run_out_off[r];
    int64_t len = run_len[r];
    if (run_kind[r] == 0) {
      uint32_t v = (uint32_t)run_value[r];
      for (int64_t i = lane; i < len; i += 64) out[off + i] = v;
    } else {
      int64_t bit0 = run_bitoff[r];
      for (int64_t i = lane; i < len; i += 64) {
        int64_t bit = bit0 + i * bit_width;
        int64_t byte = bit >> 3;
        int shift = (int)(bit & 7);
        // little-endian bit order; value spans at most 5 bytes
        uint64_t word = (uint64_t)src[byte] |
                        ((uint64_t)src[byte + 1] << 8) |
                        ((uint64_t)src[byte + 2] << 16) |
                        ((uint64_t)src[byte + 3] << 24) |
                        ((uint64_t)src[byte + 4] << 32);
        out[off + i] = (uint32_t)(word >> shift) & mask;
      }
    }
  }
}

void rle_decode_indices(const uint8_t* src, const int64_t* run_kind,
                        const int64_t* run_out_off,
                        const int64_t* run_len,
                        const int64_t* run_payload_bitoff,
                        const int64_t* run_value, int64_t n_runs,
                        int bit_width, uint32_t* out, int64_t n_out,
                        hipStream_t stream) {
  if (n_runs == 0 || n_out == 0) return;
  int64_t waves_wanted = n_runs < 16384 ? n_runs : 16384;
  int g = (int)cdiv(waves_wanted * 64, THREADS);
  if (g < 1) g = 1;
  if (g > 4096) g = 4096;
  hipLaunchKernelGGL(k_rle_decode, dim3(g), dim3(THREADS), 0, stream, src,
                     run_kind, run_out_off, run_len, run_payload_bitoff,
                     run_value, n_runs, bit_width, out, n_out);
}

// ---------------------------------------------------------------------------
// K1: parquet PLAIN page decode = unaligned-source device copy
// ---------------------------------------------------------------------------

__global__ void k_copy_aligned32(const uint32_t* __restrict__ src,
                                 uint32_t* __restrict__ dst, int64_t n32) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n32;
       i += stride)
    dst[i] = src[i];
}

__global__ void k_copy_unaligned32(const uint32_t* __restrict__ src_base,
                                   int shift_bits,
                                   uint32_t* __restrict__ dst, int64_t n32) {
  // dst word j = bytes src_base[j..j+1] funnel-shifted by the source
  // misalignment (shift_bits = 8 * (src_off & 3))
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n32;
       i += stride) {
    uint32_t lo = src_base[i];
    uint32_t hi = src_base[i + 1];
    dst[i] = (lo >> shift_bits) | (hi << (32 - shift_bits));
  }
}

void copy_unaligned(const uint8_t* src, int64_t src_off, uint8_t* dst,
                    int64_t dst_off, int64_t nbytes, hipStream_t stream) {
  if (nbytes == 0) return;
  uint32_t* d = (uint32_t*)(dst + dst_off);
  int64_t n32 = nbytes / 4;
  int a = (int)(src_off & 3);
  const uint32_t* s = (const uint32_t*)(src + (src_off & ~3ll));
  int g = grid_for(n32);
  if (a == 0) {
    hipLaunchKernelGGL(k_copy_aligned32, dim3(g), dim3(THREADS), 0, stream,
                       s, d, n32);
  } else {
    hipLaunchKernelGGL(k_copy_unaligned32, dim3(g), dim3(THREADS), 0,
                       stream, s, 8 * a, d, n32);
  }
}

// ---------------------------------------------------------------------------
// K1: snappy raw-block decompression (parquet page codec).
// One 64-lane wave per page: every lane walks the tag stream in
// lockstep (identical control flow; the redundant tag loads hit L1) and
// each op's byte movement distributes across the lanes.  Overlapping
// back-references (off < len) replicate the off-length pattern via an
// index modulo, which is byte-for-byte what serial snappy produces.
// Before a copy op reads earlier OUTPUT bytes, __syncthreads() drains
// the wave's pending vector writes (vmcnt) so other lanes' bytes are
// visible.  Pages are independent streams, so the scan's parallelism
// is pages x files.  Format: github.com/google/snappy format_description.
// ---------------------------------------------------------------------------

// Register-window stream cursor: the tag walk is inherently serial, so
// its cost is per-op DEPENDENT loads.  A 16-byte aligned window turns
// ~3-6 dependent byte loads per op into one aligned u64 load per 8
// stream bytes.
struct SnapCursor {
  const uint8_t* s0;    // aligned-down stream base
  const uint8_t* send;  // one past last valid byte
  int64_t pos;          // absolute byte position (s0-relative)
  int64_t base;         // window start (8-aligned, s0-relative)
  uint64_t lo, hi;      // 16-byte window [base, base+16)

  __device__ __forceinline__ uint64_t load8(int64_t b) {
    // aligned u64 within bounds -> single load; tail -> clamped bytes
    if (s0 + b + 8 <= send) return *(const uint64_t*)(s0 + b);
    uint64_t v = 0;
    for (int i = 0; i < 8; i++)
      if (s0 + b + i < send) v |= (uint64_t)s0[b + i] << (8 * i);
    return v;
  }
  __device__ __forceinline__ void init(const uint8_t* start,
                                       const uint8_t* end) {
    s0 = (const uint8_t*)((uintptr_t)start & ~7ull);
    send = end;
    pos = start - s0;
    base = 0;
    lo = load8(0);
    hi = load8(8);
  }
  __device__ __forceinline__ void slide() {
    while (pos - base >= 8) {
      if (pos - base >= 16) {  // long jump (literal): reinit window
        base = pos & ~7ll;
        lo = load8(base);
        hi = load8(base + 8);
        return;
      }
      lo = hi;
      base += 8;
      hi = load8(base + 8);
    }
  }
  // peek up to 8 bytes at pos (little-endian); window invariant
  // pos-base <= 7 guarantees 9+ valid bytes ahead
  __device__ __forceinline__ uint64_t peek() {
    int sh = (int)(pos - base) * 8;
    uint64_t v = lo >> sh;
    if (sh) v |= hi << (64 - sh);
    return v;
  }
  __device__ __forceinline__ void advance(int64_t nbytes) {
    pos += nbytes;
    slide();
  }
  __device__ __forceinline__ bool exhausted() {
    return s0 + pos >= send;
  }
};

__global__ void k_snappy_decomp(const uint8_t* __restrict__ src,
                                const int64_t* __restrict__ src_off,
                                const int64_t* __restrict__ src_end,
                                uint8_t* __restrict__ dst,
                                const int64_t* __restrict__ dst_off,
                                const int64_t* __restrict__ dst_len,
                                int32_t* __restrict__ status) {
  int p = blockIdx.x;
  const uint8_t* sbeg = src + src_off[p];
  const uint8_t* send = src + src_end[p];
  uint8_t* d = dst + dst_off[p];
  int64_t expected = dst_len[p];
  int lane = threadIdx.x;

  SnapCursor c;
  c.init(sbeg, send);

  // varint uncompressed length (lockstep on every lane)
  uint64_t ulen = 0;
  int shift = 0;
  while (!c.exhausted()) {
    uint8_t b = (uint8_t)c.peek();
    c.advance(1);
    ulen |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  if ((int64_t)ulen != expected) {
    if (lane == 0) status[p] = 1;
    return;
  }

  int64_t out = 0;
  int64_t synced = 0;  // output below this point is visible to all lanes
  while (out < expected && !c.exhausted()) {
    uint64_t w = c.peek();
    uint8_t tag = (uint8_t)w;
    int k = tag & 3;
    int64_t len;
    int64_t off = 0;
    const uint8_t* lit = nullptr;
    if (k == 0) {  // literal
      int64_t l = tag >> 2;
      if (l < 60) {
        len = l + 1;
        c.advance(1);
      } else {
        int nb = (int)(l - 59);  // 1..4 length bytes, little-endian
        uint32_t v = (uint32_t)((w >> 8) &
                                (0xFFFFFFFFu >> (8 * (4 - nb))));
        c.advance(1 + nb);
        len = (int64_t)v + 1;
      }
      lit = c.s0 + c.pos;
      if (lit + len > send) { if (lane == 0) status[p] = 2; return; }
      c.advance(len);
    } else if (k == 1) {  // copy, 1-byte offset
      len = ((tag >> 2) & 7) + 4;
      off = ((int64_t)(tag >> 5) << 8) | ((w >> 8) & 0xFF);
      c.advance(2);
    } else if (k == 2) {  // copy, 2-byte offset
      len = (tag >> 2) + 1;
      off = (int64_t)((w >> 8) & 0xFFFF);
      c.advance(3);
    } else {  // copy, 4-byte offset
      len = (tag >> 2) + 1;
      off = (int64_t)((w >> 8) & 0xFFFFFFFFull);
      c.advance(5);
    }
    if (out + len > expected || (k != 0 && (off <= 0 || off > out))) {
      if (lane == 0) status[p] = 3;
      return;
    }
    if (k == 0) {
      // wide literal copy: 16B lanes over the dest-aligned middle
      // (incompressible parquet pages — e.g. random doubles — arrive
      // as ONE giant literal, so this path carries ~half the bytes)
      if (len >= 256) {
        int64_t head = (16 - ((uintptr_t)(d + out) & 15)) & 15;
        int64_t body = (len - head) & ~15ll;
        for (int64_t i = lane; i < head; i += 64) d[out + i] = lit[i];
        const uint8_t* lsrc = lit + head;
        uint8_t* ldst = d + out + head;
        if (lsrc + body + 16 <= send) {
          for (int64_t i = lane; i * 16 < body; i += 64) {
            uint4 v;
            __builtin_memcpy(&v, lsrc + i * 16, 16);
            *(uint4*)(ldst + i * 16) = v;
          }
        } else {
          for (int64_t i = lane; i < body; i += 64) ldst[i] = lsrc[i];
        }
        for (int64_t i = head + body + lane; i < len; i += 64)
          d[out + i] = lit[i];
      } else {
        for (int64_t i = lane; i < len; i += 64) d[out + i] = lit[i];
      }
    } else {
      // drain this wave's pending stores only when the source range
      // reaches past the last drain point (could include other lanes'
      // recent bytes).  One wave per block: a full barrier is not
      // needed — s_waitcnt vmcnt(0) makes every lane's stores visible
      // through the CU's L1 to the wave's subsequent loads.
      if (out - off + len > synced) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        synced = out;
      }
      if (off >= len) {
        if (len >= 256 && off >= 8) {
          // long disjoint copy: 8B-granule, dest-aligned middle (short
          // copies — the common case — stay on the 1-instruction
          // byte path; the wide path only pays off past ~4 B/lane)
          int64_t head = (8 - ((uintptr_t)(d + out) & 7)) & 7;
          if (head > len) head = len;
          int64_t body = (len - head) & ~7ll;
          for (int64_t i = lane; i < head; i += 64)
            d[out + i] = d[out - off + i];
          const uint8_t* csrc = d + out - off + head;
          uint8_t* cdst = d + out + head;
          for (int64_t i = lane; i * 8 < body; i += 64) {
            uint64_t v;
            __builtin_memcpy(&v, csrc + i * 8, 8);
            *(uint64_t*)(cdst + i * 8) = v;
          }
          for (int64_t i = head + body + lane; i < len; i += 64)
            d[out + i] = d[out - off + i];
        } else {
          for (int64_t i = lane; i < len; i += 64)
            d[out + i] = d[out - off + i];
        }
      } else {
        for (int64_t i = lane; i < len; i += 64)
          d[out + i] = d[out - off + (i % off)];
      }
    }
    out += len;
  }
  if (lane == 0) status[p] = out == expected ? 0 : 4;
}

void snappy_decompress_pages(const uint8_t* src, const int64_t* src_off,
                             const int64_t* src_end, uint8_t* dst,
                             const int64_t* dst_off,
                             const int64_t* dst_len, int32_t* status,
                             int n_pages, hipStream_t stream) {
  if (n_pages == 0) return;
  hipLaunchKernelGGL(k_snappy_decomp, dim3(n_pages), dim3(64), 0, stream,
                     src, src_off, src_end, dst, dst_off, dst_len,
                     status);
}

}  // namespace hsk
