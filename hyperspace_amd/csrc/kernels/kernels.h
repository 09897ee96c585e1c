// Host-side API of the CDNA4 kernel suite (pure HIP, gfx950).
// Implemented in kernels.hip; called from the torch binding (binding.cpp).
#pragma once

#include <cstdint>
#include <hip/hip_runtime.h>

namespace hsk {

// K2: Spark-compatible Murmur3 fold over one column into h (u32 in u32 buf).
// kind: 0 = i32-hash (int32 input), 1 = i64-hash (int64 input).
// valid: optional per-row validity bytes — a null row leaves h unchanged
// (Spark Murmur3Hash skips null children); nullptr = all valid.
void murmur3_column(const void* vals, int kind, const uint8_t* valid,
                    uint32_t* h, int64_t n, bool first, uint32_t seed,
                    hipStream_t stream);
void pmod_buckets(const uint32_t* h, int32_t* out, int64_t n,
                  int32_t num_buckets, hipStream_t stream);

// order-preserving u64 normalization (dtype codes: 0=i64,1=i32,2=f64,3=f32,
// 4=i16, 5=i8, 6=bool/u8)
void normalize_key(const void* vals, int dtype, uint64_t* out, int64_t n,
                   hipStream_t stream);

// K3: stable LSD radix sort of (u64 key, i64 payload), ascending u64 order.
// keys/payload are sorted in place; tmp buffers same size provided by caller.
void radix_sort_pairs(uint64_t* keys, int64_t* payload, uint64_t* tmp_keys,
                      int64_t* tmp_payload, uint32_t* hist, uint64_t* d_mask,
                      int64_t n, hipStream_t stream);
void radix_sort_pairs32(uint64_t* keys, int32_t* payload,
                        uint64_t* tmp_keys, int32_t* tmp_payload,
                        uint32_t* hist, uint64_t* d_mask, int64_t n,
                        hipStream_t stream);
// histogram buffer size requirement (u32 elements)
int64_t radix_sort_hist_size(int64_t n);

// K4: segmented sorted merge join, two-phase per-tile form.  Phase 1
// writes one pair-count per 256-row left tile; after an exclusive scan
// of tile counts, phase 2 recomputes the tile-narrowed searches and
// emits pairs at tile_offset + intra-tile LDS prefix (no per-row
// metadata arrays touch HBM).
int64_t merge_join_tile_size();
// K4b: segmented two-sorted-run merge permutation (split[b] = A/B
// boundary per bucket; == seg[b+1] for single-run buckets).
void run_merge_perm(const uint64_t* keys, const int64_t* seg,
                    const int64_t* split, int64_t n, int64_t n_seg,
                    int64_t* perm, hipStream_t stream);
void merge_join_tile_count(const uint64_t* lkeys, const uint64_t* rkeys,
                           const int64_t* lseg, const int64_t* rseg,
                           int64_t n_left, int64_t n_seg,
                           int64_t* tile_counts, hipStream_t stream);
void merge_join_tile_emit(const uint64_t* lkeys, const uint64_t* rkeys,
                          const int64_t* lseg, const int64_t* rseg,
                          int64_t n_left, int64_t n_seg,
                          const int64_t* tile_offsets, int64_t* out_l,
                          int64_t* out_r, hipStream_t stream);

// generic exclusive scan over i64 (single pass, device-wide)
void exclusive_scan_i64(const int64_t* in, int64_t* out, int64_t n,
                        int64_t* total, hipStream_t stream);

// filter scan: indices i (ascending) with lo <= key[i] <= hi (u64 order,
// inclusive flags).  Two-phase: count (fills block_counts, scans them in
// place, writes total) then emit (stable compaction using the scanned
// block bases).  block_counts sized select_num_blocks(n).
int64_t select_num_blocks(int64_t n);
void select_range_count(const uint64_t* keys, int64_t n, uint64_t lo,
                        uint64_t hi, bool lo_incl, bool hi_incl,
                        int64_t* block_counts, int64_t* total,
                        hipStream_t stream);
void select_range_emit(const uint64_t* keys, int64_t n, uint64_t lo,
                       uint64_t hi, bool lo_incl, bool hi_incl,
                       const int64_t* block_bases, int64_t* out_idx,
                       hipStream_t stream);

// K7: membership of values in a sorted set -> bool mask
void isin_sorted(const int64_t* vals, int64_t n, const int64_t* sorted_set,
                 int64_t m, bool* out, hipStream_t stream);

// K8: per-segment min/max of int64 values
void segmented_minmax(const int64_t* vals, const int64_t* seg_off,
                      int64_t n_seg, int64_t* mins, int64_t* maxs,
                      hipStream_t stream);

// K8: bloom filter build/probe (k hashes, m_bits bits over u64 words)
void bloom_build(const int64_t* vals, int64_t n, uint64_t* words,
                 int64_t m_bits, int k, hipStream_t stream);
void bloom_probe(const int64_t* vals, int64_t n, const uint64_t* words,
                 int64_t m_bits, int k, bool* out, hipStream_t stream);
// K9: batched device sketch-predicate probe — every value against every
// per-file filter; out[f] |= filter f may contain any value
void bloom_probe_many(const int64_t* vals, int64_t n_vals,
                      const uint64_t* words, int64_t words_per_filter,
                      int64_t n_filters, int64_t m_bits, int k, bool* out,
                      hipStream_t stream);

// K10: z-order bit interleave of up to 8 normalized u64 columns
void zorder_key(const uint64_t* const* cols, int n_cols, int bits_per_col,
                int64_t n, uint64_t* out, hipStream_t stream);

// row gather: out[i] = in[idx[i]] for element sizes 4/8
void gather(const void* in, const int64_t* idx, void* out, int64_t n,
            int elem_size, hipStream_t stream);

// K1 parquet dictionary-page decode: expand an RLE/bit-packed hybrid
// run table (host-parsed) into u32 dictionary indices.
// runs arrays (one entry per run):
//   kind: 0 = repeated (value in run_value), 1 = bit-packed literal
//         (bits start at src bit offset run_payload_bitoff)
//   out_off: first output index of the run; len: output count
void rle_decode_indices(const uint8_t* src, const int64_t* run_kind,
                        const int64_t* run_out_off,
                        const int64_t* run_len,
                        const int64_t* run_payload_bitoff,
                        const int64_t* run_value, int64_t n_runs,
                        int bit_width, uint32_t* out, int64_t n_out,
                        hipStream_t stream);

// K1 parquet page decode: copy nbytes from (src + src_off) — any byte
// alignment — to 4-byte-aligned dst+dst_off.  nbytes % 4 == 0.  The src
// buffer must extend >= 4 bytes past src_off+nbytes (parquet footers
// guarantee this for page payloads).
// K1 snappy raw-block page decompression: one wave per page; status[p]
// = 0 on success, nonzero on malformed input (caller falls back).
void snappy_decompress_pages(const uint8_t* src, const int64_t* src_off,
                             const int64_t* src_end, uint8_t* dst,
                             const int64_t* dst_off,
                             const int64_t* dst_len, int32_t* status,
                             int n_pages, hipStream_t stream);

void copy_unaligned(const uint8_t* src, int64_t src_off, uint8_t* dst,
                    int64_t dst_off, int64_t nbytes, hipStream_t stream);

}  // namespace hsk
