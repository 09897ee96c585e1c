// Torch extension binding for the CDNA4 kernel suite (kernels.hip).
// Written directly against the ROCm torch C++ surface (c10::hip) — no
// CUDA-compat naming, no hipify.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

#include <algorithm>
#include <tuple>
#include <vector>

#include "kernels/kernels.h"

namespace {

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_cuda(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a device tensor");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

int normalize_dtype_code(torch::ScalarType t) {
  switch (t) {
    case torch::kInt64: return 0;
    case torch::kInt32: return 1;
    case torch::kFloat64: return 2;
    case torch::kFloat32: return 3;
    case torch::kInt16: return 4;
    case torch::kInt8: return 5;
    case torch::kBool: return 6;
    default:
      TORCH_CHECK(false, "unsupported sort-key dtype");
  }
}

torch::Tensor murmur3_bucket(std::vector<torch::Tensor> keys,
                             int64_t num_buckets,
                             std::vector<torch::Tensor> masks = {}) {
  TORCH_CHECK(!keys.empty(), "need at least one key column");
  TORCH_CHECK(masks.empty() || masks.size() == keys.size(),
              "masks must be empty or parallel to keys");
  auto n = keys[0].numel();
  auto dev = keys[0].device();
  auto h = torch::empty({n}, torch::dtype(torch::kInt32).device(dev));
  auto stream = current_stream();
  bool first = true;
  for (size_t ci = 0; ci < keys.size(); ++ci) {
    auto& k = keys[ci];
    check_cuda(k, "key");
    TORCH_CHECK(k.numel() == n, "key column length mismatch");
    torch::Tensor col = k;
    int kind;
    switch (col.scalar_type()) {
      case torch::kInt64:
      case torch::kFloat64:
        kind = 1;
        break;
      case torch::kInt32:
      case torch::kFloat32:
        kind = 0;
        break;
      case torch::kInt16:
      case torch::kInt8:
      case torch::kBool:
        col = col.to(torch::kInt32);
        kind = 0;
        break;
      default:
        TORCH_CHECK(false, "unsupported hash-key dtype");
    }
    // an empty mask tensor means "column has no nulls"
    const uint8_t* valid = nullptr;
    torch::Tensor mcol;
    if (!masks.empty() && masks[ci].numel() > 0) {
      mcol = masks[ci];
      check_cuda(mcol, "mask");
      TORCH_CHECK(mcol.numel() == n, "mask length mismatch");
      TORCH_CHECK(mcol.scalar_type() == torch::kBool ||
                      mcol.scalar_type() == torch::kUInt8,
                  "mask must be bool/uint8");
      valid = (const uint8_t*)mcol.data_ptr();
    }
    hsk::murmur3_column(col.data_ptr(), kind, valid,
                        (uint32_t*)h.data_ptr<int32_t>(), n, first,
                        /*seed=*/42u, stream);
    first = false;
  }
  auto out = torch::empty({n}, torch::dtype(torch::kInt32).device(dev));
  hsk::pmod_buckets((const uint32_t*)h.data_ptr<int32_t>(),
                    out.data_ptr<int32_t>(), n, (int32_t)num_buckets,
                    stream);
  return out;
}

torch::Tensor normalize_key(torch::Tensor vals) {
  check_cuda(vals, "vals");
  auto n = vals.numel();
  auto out = torch::empty(
      {n}, torch::dtype(torch::kInt64).device(vals.device()));
  hsk::normalize_key(vals.data_ptr(),
                     normalize_dtype_code(vals.scalar_type()),
                     (uint64_t*)out.data_ptr<int64_t>(), n,
                     current_stream());
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> radix_sort_pairs(
    torch::Tensor keys, torch::Tensor payload) {
  check_cuda(keys, "keys");
  check_cuda(payload, "payload");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  TORCH_CHECK(payload.scalar_type() == torch::kInt64,
              "payload must be int64");
  auto n = keys.numel();
  auto out_keys = keys.clone();
  if (n <= 1) return {out_keys, payload.clone()};
  auto hist = torch::empty({hsk::radix_sort_hist_size(n)},
                           torch::dtype(torch::kInt32).device(keys.device()));
  auto mask = torch::empty({1},
                           torch::dtype(torch::kInt64).device(keys.device()));
  auto tmp_keys = torch::empty_like(keys);
  // payload values are row indices in every engine call site: when they
  // fit int32 the sort runs the narrow-payload pipeline (2/3 the
  // per-pass traffic) and widens at the end
  bool fits32 = n < (int64_t)INT32_MAX;
  if (fits32) {
    auto mm = payload.aminmax();
    fits32 = std::get<0>(mm).item<int64_t>() >= 0 &&
             std::get<1>(mm).item<int64_t>() < (int64_t)INT32_MAX;
  }
  if (fits32) {
    auto p32 = payload.to(torch::kInt32);
    auto tmp_p32 = torch::empty_like(p32);
    hsk::radix_sort_pairs32((uint64_t*)out_keys.data_ptr<int64_t>(),
                            p32.data_ptr<int32_t>(),
                            (uint64_t*)tmp_keys.data_ptr<int64_t>(),
                            tmp_p32.data_ptr<int32_t>(),
                            (uint32_t*)hist.data_ptr<int32_t>(),
                            (uint64_t*)mask.data_ptr<int64_t>(), n,
                            current_stream());
    return {out_keys, p32.to(torch::kInt64)};
  }
  auto out_payload = payload.clone();
  auto tmp_payload = torch::empty_like(payload);
  hsk::radix_sort_pairs((uint64_t*)out_keys.data_ptr<int64_t>(),
                        out_payload.data_ptr<int64_t>(),
                        (uint64_t*)tmp_keys.data_ptr<int64_t>(),
                        tmp_payload.data_ptr<int64_t>(),
                        (uint32_t*)hist.data_ptr<int32_t>(),
                        (uint64_t*)mask.data_ptr<int64_t>(), n,
                        current_stream());
  return {out_keys, out_payload};
}

std::tuple<torch::Tensor, torch::Tensor> merge_join(torch::Tensor lkeys,
                                                    torch::Tensor rkeys,
                                                    torch::Tensor lseg,
                                                    torch::Tensor rseg) {
  check_cuda(lkeys, "lkeys");
  check_cuda(rkeys, "rkeys");
  auto dev = lkeys.device();
  auto opts = torch::dtype(torch::kInt64).device(dev);
  // segment offsets may arrive on CPU
  auto lseg_d = lseg.to(dev, torch::kInt64).contiguous();
  auto rseg_d = rseg.to(dev, torch::kInt64).contiguous();
  int64_t n_left = lkeys.numel();
  int64_t n_seg = lseg_d.numel() - 1;
  auto stream = current_stream();
  if (n_left == 0 || rkeys.numel() == 0) {
    return {torch::empty({0}, opts), torch::empty({0}, opts)};
  }
  int64_t tile = hsk::merge_join_tile_size();
  int64_t n_tiles = (n_left + tile - 1) / tile;
  auto tile_counts = torch::empty({n_tiles}, opts);
  hsk::merge_join_tile_count((const uint64_t*)lkeys.data_ptr<int64_t>(),
                             (const uint64_t*)rkeys.data_ptr<int64_t>(),
                             lseg_d.data_ptr<int64_t>(),
                             rseg_d.data_ptr<int64_t>(), n_left, n_seg,
                             tile_counts.data_ptr<int64_t>(), stream);
  auto tile_offsets = torch::empty({n_tiles}, opts);
  auto total = torch::empty({1}, opts);
  hsk::exclusive_scan_i64(tile_counts.data_ptr<int64_t>(),
                          tile_offsets.data_ptr<int64_t>(), n_tiles,
                          total.data_ptr<int64_t>(), stream);
  int64_t n_out = total.cpu().item<int64_t>();
  auto out_l = torch::empty({n_out}, opts);
  auto out_r = torch::empty({n_out}, opts);
  if (n_out > 0) {
    hsk::merge_join_tile_emit((const uint64_t*)lkeys.data_ptr<int64_t>(),
                              (const uint64_t*)rkeys.data_ptr<int64_t>(),
                              lseg_d.data_ptr<int64_t>(),
                              rseg_d.data_ptr<int64_t>(), n_left, n_seg,
                              tile_offsets.data_ptr<int64_t>(),
                              out_l.data_ptr<int64_t>(),
                              out_r.data_ptr<int64_t>(), stream);
  }
  return {out_l, out_r};
}

torch::Tensor isin_sorted(torch::Tensor vals, torch::Tensor sorted_set) {
  check_cuda(vals, "vals");
  auto n = vals.numel();
  auto out = torch::empty(
      {n}, torch::dtype(torch::kBool).device(vals.device()));
  if (sorted_set.numel() == 0) {
    out.zero_();
    return out;
  }
  auto set_d = sorted_set.to(vals.device(), torch::kInt64).contiguous();
  hsk::isin_sorted(vals.data_ptr<int64_t>(), n, set_d.data_ptr<int64_t>(),
                   set_d.numel(), out.data_ptr<bool>(), current_stream());
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> segmented_minmax(
    torch::Tensor vals, torch::Tensor seg_off) {
  check_cuda(vals, "vals");
  auto seg_d = seg_off.to(vals.device(), torch::kInt64).contiguous();
  int64_t n_seg = seg_d.numel() - 1;
  auto opts = torch::dtype(torch::kInt64).device(vals.device());
  auto mins = torch::empty({n_seg}, opts);
  auto maxs = torch::empty({n_seg}, opts);
  hsk::segmented_minmax(vals.data_ptr<int64_t>(), seg_d.data_ptr<int64_t>(),
                        n_seg, mins.data_ptr<int64_t>(),
                        maxs.data_ptr<int64_t>(), current_stream());
  return {mins, maxs};
}

torch::Tensor bloom_build(torch::Tensor vals, int64_t m_bits, int64_t k) {
  check_cuda(vals, "vals");
  int64_t words = (m_bits + 63) / 64;
  auto out = torch::zeros(
      {words}, torch::dtype(torch::kInt64).device(vals.device()));
  hsk::bloom_build(vals.data_ptr<int64_t>(), vals.numel(),
                   (uint64_t*)out.data_ptr<int64_t>(), m_bits, (int)k,
                   current_stream());
  return out;
}

torch::Tensor bloom_probe(torch::Tensor vals, torch::Tensor words,
                          int64_t m_bits, int64_t k) {
  check_cuda(vals, "vals");
  auto words_d = words.to(vals.device(), torch::kInt64).contiguous();
  auto out = torch::empty(
      {vals.numel()}, torch::dtype(torch::kBool).device(vals.device()));
  hsk::bloom_probe(vals.data_ptr<int64_t>(), vals.numel(),
                   (const uint64_t*)words_d.data_ptr<int64_t>(), m_bits,
                   (int)k, out.data_ptr<bool>(), current_stream());
  return out;
}

torch::Tensor bloom_probe_many(torch::Tensor vals, torch::Tensor words,
                               int64_t m_bits, int64_t k) {
  check_cuda(vals, "vals");
  TORCH_CHECK(words.dim() == 2, "words must be [n_filters, words]");
  auto words_d = words.to(vals.device(), torch::kInt64).contiguous();
  int64_t n_filters = words_d.size(0);
  auto out = torch::zeros(
      {n_filters}, torch::dtype(torch::kBool).device(vals.device()));
  hsk::bloom_probe_many(vals.data_ptr<int64_t>(), vals.numel(),
                        (const uint64_t*)words_d.data_ptr<int64_t>(),
                        words_d.size(1), n_filters, m_bits, (int)k,
                        out.data_ptr<bool>(), current_stream());
  return out;
}

torch::Tensor zorder_key(std::vector<torch::Tensor> cols,
                         int64_t bits_per_col) {
  TORCH_CHECK(!cols.empty() && cols.size() <= 8, "1..8 zorder columns");
  auto n = cols[0].numel();
  std::vector<const uint64_t*> ptrs;
  for (auto& c : cols) {
    check_cuda(c, "zorder col");
    ptrs.push_back((const uint64_t*)c.data_ptr<int64_t>());
  }
  auto out = torch::empty(
      {n}, torch::dtype(torch::kInt64).device(cols[0].device()));
  hsk::zorder_key(ptrs.data(), (int)ptrs.size(), (int)bits_per_col, n,
                  (uint64_t*)out.data_ptr<int64_t>(), current_stream());
  return out;
}

torch::Tensor run_merge_perm(torch::Tensor keys, torch::Tensor seg,
                             torch::Tensor split) {
  check_cuda(keys, "keys");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  auto dev = keys.device();
  auto i64 = torch::dtype(torch::kInt64);
  auto seg_d = seg.to(dev, torch::kInt64).contiguous();
  auto split_d = split.to(dev, torch::kInt64).contiguous();
  int64_t n = keys.numel();
  int64_t n_seg = seg_d.numel() - 1;
  TORCH_CHECK(split_d.numel() == n_seg, "split must have n_seg entries");
  auto perm = torch::empty({n}, i64.device(dev));
  hsk::run_merge_perm((const uint64_t*)keys.data_ptr<int64_t>(),
                      seg_d.data_ptr<int64_t>(),
                      split_d.data_ptr<int64_t>(), n, n_seg,
                      perm.data_ptr<int64_t>(), current_stream());
  return perm;
}

torch::Tensor snappy_decompress(torch::Tensor dev_bytes,
                                torch::Tensor src_off,
                                torch::Tensor src_end, torch::Tensor dst,
                                torch::Tensor dst_off,
                                torch::Tensor dst_len) {
  check_cuda(dev_bytes, "dev_bytes");
  check_cuda(dst, "dst");
  TORCH_CHECK(dev_bytes.scalar_type() == torch::kUInt8 &&
                  dst.scalar_type() == torch::kUInt8,
              "byte tensors required");
  int64_t n_pages = src_off.numel();
  auto dev = dev_bytes.device();
  auto i64 = torch::dtype(torch::kInt64).device(dev);
  auto so = src_off.to(i64.device(), torch::kInt64).contiguous();
  auto se = src_end.to(i64.device(), torch::kInt64).contiguous();
  auto dofs = dst_off.to(i64.device(), torch::kInt64).contiguous();
  auto dl = dst_len.to(i64.device(), torch::kInt64).contiguous();
  auto status = torch::zeros(
      {n_pages}, torch::dtype(torch::kInt32).device(dev));
  hsk::snappy_decompress_pages(
      dev_bytes.data_ptr<uint8_t>(), so.data_ptr<int64_t>(),
      se.data_ptr<int64_t>(), dst.data_ptr<uint8_t>(),
      dofs.data_ptr<int64_t>(), dl.data_ptr<int64_t>(),
      status.data_ptr<int32_t>(), (int)n_pages, current_stream());
  return status;
}

// Host decode of RLE-hybrid definition levels (max_def = 1) into a
// validity byte array: the per-page Python decoder costs ~1 ms/page on
// pyarrow's many-run encodings; this tight loop is ~20 us.
torch::Tensor decode_def_levels(py::buffer data, int64_t off,
                                int64_t length, int64_t n) {
  py::buffer_info info = data.request();
  const uint8_t* base = (const uint8_t*)info.ptr;
  TORCH_CHECK(off >= 0 && off + length <= (int64_t)info.size,
              "def-levels out of range");
  const uint8_t* p = base + off;
  const uint8_t* end = p + length;
  auto out = torch::empty({n}, torch::dtype(torch::kBool));
  bool* o = out.data_ptr<bool>();
  int64_t filled = 0;
  while (filled < n && p < end) {
    uint64_t header = 0;
    int shift = 0;
    while (true) {
      TORCH_CHECK(p < end, "def-levels: truncated varint");
      uint8_t b = *p++;
      header |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    if (header & 1) {  // bit-packed run: groups of 8 levels
      int64_t groups = (int64_t)(header >> 1);
      int64_t cnt = groups * 8;
      if (cnt > n - filled) cnt = n - filled;
      TORCH_CHECK(p + groups <= end, "def-levels: truncated bitpack");
      for (int64_t i = 0; i < cnt; ++i)
        o[filled + i] = (p[i >> 3] >> (i & 7)) & 1;
      p += groups;
      filled += cnt;
    } else {  // RLE run: one value byte at bit width 1
      int64_t run = (int64_t)(header >> 1);
      if (run > n - filled) run = n - filled;
      TORCH_CHECK(p < end, "def-levels: truncated rle value");
      bool v = *p++ != 0;
      std::fill(o + filled, o + filled + run, v);
      filled += run;
    }
  }
  std::fill(o + filled, o + n, true);
  return out;
}

// Batched RLE-hybrid run parse over MANY pages in one GIL-released
// call: the per-page python loop cost dominated multi-row-group snappy
// chunk decode (hundreds of pages x ~0.5 ms of interpreter work).
// Emits runs already shifted into the caller's output-row and
// bit-offset coordinate spaces, plus per-page run counts.
std::vector<torch::Tensor> parse_rle_runs_batch(
    torch::Tensor bytes, torch::Tensor starts, torch::Tensor ends,
    torch::Tensor bws, torch::Tensor n_valids, torch::Tensor out_shifts,
    torch::Tensor bit_shifts) {
  TORCH_CHECK(!bytes.is_cuda() && bytes.scalar_type() == torch::kUInt8,
              "bytes must be a cpu u8 tensor");
  const uint8_t* d = bytes.data_ptr<uint8_t>();
  int64_t n_pages = starts.numel();
  auto sa = starts.data_ptr<int64_t>();
  auto ea = ends.data_ptr<int64_t>();
  auto ba = bws.data_ptr<int64_t>();
  auto va = n_valids.data_ptr<int64_t>();
  auto oa = out_shifts.data_ptr<int64_t>();
  auto fa = bit_shifts.data_ptr<int64_t>();
  std::vector<int64_t> kind, out_off, len, bitoff, value, counts;
  counts.reserve(n_pages);
  for (int64_t pg = 0; pg < n_pages; ++pg) {
    int64_t pos = sa[pg], end = ea[pg], out = 0;
    int64_t bit_width = ba[pg], num_values = va[pg];
    int64_t bw_bytes = (bit_width + 7) / 8;
    int64_t n_runs0 = (int64_t)kind.size();
    while (out < num_values && pos < end) {
      uint64_t h = 0;
      int shift = 0;
      while (true) {
        TORCH_CHECK(pos < end, "rle: truncated varint");
        uint8_t b = d[pos++];
        h |= (uint64_t)(b & 0x7F) << shift;
        if (!(b & 0x80)) break;
        shift += 7;
      }
      if (h & 1) {
        int64_t groups = (int64_t)(h >> 1);
        int64_t count = groups * 8;
        if (count > num_values - out) count = num_values - out;
        kind.push_back(1);
        out_off.push_back(out + oa[pg]);
        len.push_back(count);
        bitoff.push_back(pos * 8 + fa[pg]);
        value.push_back(0);
        pos += groups * bit_width;
        out += count;
      } else {
        int64_t run = (int64_t)(h >> 1);
        if (run > num_values - out) run = num_values - out;
        int64_t v = 0;
        for (int64_t i = 0; i < bw_bytes; i++) {
          TORCH_CHECK(pos < end, "rle: truncated repeat value");
          v |= (int64_t)d[pos++] << (8 * i);
        }
        kind.push_back(0);
        out_off.push_back(out + oa[pg]);
        len.push_back(run);
        bitoff.push_back(0);
        value.push_back(v);
        out += run;
      }
    }
    TORCH_CHECK(out == num_values, "rle: page ", pg, " produced ", out,
                " of ", num_values, " values");
    counts.push_back((int64_t)kind.size() - n_runs0);
  }
  auto opts = torch::dtype(torch::kInt64);
  auto mk = [&](std::vector<int64_t>& v) { return torch::tensor(v, opts); };
  return {mk(kind), mk(out_off), mk(len), mk(bitoff), mk(value),
          mk(counts)};
}

// PLAIN BYTE_ARRAY page parse (K1 string path): each value is a 4-byte
// little-endian length prefix + payload.  Concatenates the values of
// every listed page into one arrow-style (int32 offsets, bytes) pair so
// the caller can dictionary-encode in one pass (parquet-format.md
// Encodings: PLAIN for BYTE_ARRAY; reference uses parquet-mr's reader).
std::vector<torch::Tensor> parse_byte_arrays(
    torch::Tensor bytes, torch::Tensor starts, torch::Tensor ends,
    torch::Tensor counts) {
  TORCH_CHECK(!bytes.is_cuda() && bytes.scalar_type() == torch::kUInt8,
              "bytes must be a cpu u8 tensor");
  const uint8_t* d = bytes.data_ptr<uint8_t>();
  int64_t n_pages = starts.numel();
  auto sa = starts.data_ptr<int64_t>();
  auto ea = ends.data_ptr<int64_t>();
  auto ca = counts.data_ptr<int64_t>();
  int64_t total_n = 0, cap = 0;
  for (int64_t p = 0; p < n_pages; ++p) {
    total_n += ca[p];
    cap += ea[p] - sa[p];
  }
  TORCH_CHECK(cap <= INT32_MAX, "byte-array payload exceeds int32 offsets");
  auto offs = torch::empty({total_n + 1}, torch::dtype(torch::kInt32));
  auto out = torch::empty({cap}, torch::dtype(torch::kUInt8));
  int32_t* op = offs.data_ptr<int32_t>();
  uint8_t* bp = out.data_ptr<uint8_t>();
  int64_t written = 0, vi = 0;
  for (int64_t p = 0; p < n_pages; ++p) {
    int64_t pos = sa[p], end = ea[p];
    for (int64_t k = 0; k < ca[p]; ++k) {
      TORCH_CHECK(pos + 4 <= end, "byte-array page ", p,
                  ": truncated length prefix");
      uint32_t len = (uint32_t)d[pos] | ((uint32_t)d[pos + 1] << 8) |
                     ((uint32_t)d[pos + 2] << 16) |
                     ((uint32_t)d[pos + 3] << 24);
      pos += 4;
      TORCH_CHECK(pos + (int64_t)len <= end, "byte-array page ", p,
                  ": value overruns page");
      op[vi++] = (int32_t)written;
      memcpy(bp + written, d + pos, len);
      written += len;
      pos += len;
    }
  }
  op[vi] = (int32_t)written;
  return {offs, out.narrow(0, 0, written)};
}

// GIL-released memcpy of a buffer-protocol object (e.g. a pyarrow
// Buffer from Codec::Decompress) into a cpu uint8 tensor slice — a
// plain numpy slice assignment holds the GIL for the whole copy, which
// serializes the 16-thread host-codec upload staging.
void host_memcpy(torch::Tensor dst, int64_t off, py::buffer src) {
  TORCH_CHECK(!dst.is_cuda() && dst.scalar_type() == torch::kUInt8,
              "dst must be a cpu u8 tensor");
  py::buffer_info info = src.request();
  int64_t n = (int64_t)info.size * (int64_t)info.itemsize;
  TORCH_CHECK(off >= 0 && off + n <= dst.numel(),
              "host_memcpy out of range");
  uint8_t* d = dst.data_ptr<uint8_t>() + off;
  const void* s = info.ptr;
  {
    py::gil_scoped_release rel;
    memcpy(d, s, (size_t)n);
  }
}

std::vector<torch::Tensor> parse_rle_runs(torch::Tensor bytes,
                                           int64_t start, int64_t end,
                                           int64_t bit_width,
                                           int64_t num_values) {
  TORCH_CHECK(!bytes.is_cuda() && bytes.scalar_type() == torch::kUInt8,
              "bytes must be a cpu u8 tensor");
  const uint8_t* d = bytes.data_ptr<uint8_t>();
  std::vector<int64_t> kind, out_off, len, bitoff, value;
  int64_t pos = start, out = 0;
  int64_t bw_bytes = (bit_width + 7) / 8;
  while (out < num_values && pos < end) {
    // varint header
    uint64_t h = 0;
    int shift = 0;
    while (true) {
      TORCH_CHECK(pos < end, "rle: truncated varint");
      uint8_t b = d[pos++];
      h |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    if (h & 1) {
      int64_t groups = (int64_t)(h >> 1);
      int64_t count = groups * 8;
      if (count > num_values - out) count = num_values - out;
      kind.push_back(1);
      out_off.push_back(out);
      len.push_back(count);
      bitoff.push_back(pos * 8);
      value.push_back(0);
      pos += groups * bit_width;  // payload bytes = groups * bit_width
      out += count;
    } else {
      int64_t run = (int64_t)(h >> 1);
      if (run > num_values - out) run = num_values - out;
      int64_t v = 0;
      for (int64_t i = 0; i < bw_bytes; i++) {
        TORCH_CHECK(pos < end, "rle: truncated repeat value");
        v |= (int64_t)d[pos++] << (8 * i);
      }
      kind.push_back(0);
      out_off.push_back(out);
      len.push_back(run);
      bitoff.push_back(0);
      value.push_back(v);
      out += run;
    }
  }
  TORCH_CHECK(out == num_values, "rle: produced ", out, " of ",
              num_values, " values");
  auto opts = torch::dtype(torch::kInt64);
  auto mk = [&](std::vector<int64_t>& v) {
    return torch::tensor(v, opts);
  };
  return {mk(kind), mk(out_off), mk(len), mk(bitoff), mk(value)};
}

torch::Tensor rle_decode(torch::Tensor src_u8_dev, torch::Tensor kind,
                         torch::Tensor out_off, torch::Tensor len,
                         torch::Tensor bitoff, torch::Tensor value,
                         int64_t bit_width, int64_t n_out) {
  check_cuda(src_u8_dev, "src");
  auto dev = src_u8_dev.device();
  auto k = kind.to(dev), o = out_off.to(dev), l = len.to(dev),
       b = bitoff.to(dev), v = value.to(dev);
  auto out = torch::empty({n_out},
                          torch::dtype(torch::kInt32).device(dev));
  hsk::rle_decode_indices(src_u8_dev.data_ptr<uint8_t>(),
                          k.data_ptr<int64_t>(), o.data_ptr<int64_t>(),
                          l.data_ptr<int64_t>(), b.data_ptr<int64_t>(),
                          v.data_ptr<int64_t>(), kind.numel(),
                          (int)bit_width,
                          (uint32_t*)out.data_ptr<int32_t>(), n_out,
                          current_stream());
  return out;
}

void copy_unaligned(torch::Tensor src_u8, int64_t src_off,
                    torch::Tensor dst, int64_t dst_byte_off,
                    int64_t nbytes) {
  TORCH_CHECK(src_u8.is_cuda() && dst.is_cuda(), "device tensors required");
  TORCH_CHECK(src_u8.scalar_type() == torch::kUInt8, "src must be u8");
  TORCH_CHECK(src_off + nbytes + 4 <= src_u8.numel(),
              "src buffer must extend 4 bytes past the payload");
  TORCH_CHECK(dst_byte_off % 4 == 0 && nbytes % 4 == 0, "4B granularity");
  hsk::copy_unaligned(src_u8.data_ptr<uint8_t>(), src_off,
                      (uint8_t*)dst.data_ptr(), dst_byte_off, nbytes,
                      current_stream());
}

torch::Tensor gather_rows(torch::Tensor values, torch::Tensor idx) {
  check_cuda(values, "values");
  check_cuda(idx, "idx");
  TORCH_CHECK(idx.scalar_type() == torch::kInt64, "idx must be int64");
  auto out = torch::empty({idx.numel()}, values.options());
  hsk::gather(values.data_ptr(), idx.data_ptr<int64_t>(), out.data_ptr(),
              idx.numel(), (int)values.element_size(), current_stream());
  return out;
}

}  // namespace

// out-of-namespace impl so the two-phase select is a single public symbol
torch::Tensor select_range_u64_pub(torch::Tensor keys, int64_t lo,
                                   int64_t hi, bool lo_incl, bool hi_incl) {
  TORCH_CHECK(keys.is_cuda() && keys.is_contiguous(), "keys");
  auto n = keys.numel();
  auto opts = torch::dtype(torch::kInt64).device(keys.device());
  if (n == 0) return torch::empty({0}, opts);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int64_t nb = hsk::select_num_blocks(n);
  auto bcounts = torch::empty({nb}, opts);
  auto total = torch::empty({1}, opts);
  // lo/hi arrive already u64-normalized (int64-encoded); the kernel
  // compares raw u64 bits, so reinterpret without remapping
  uint64_t ulo = (uint64_t)lo;
  uint64_t uhi = (uint64_t)hi;
  // phase 1: count + scan
  hsk::select_range_count((const uint64_t*)keys.data_ptr<int64_t>(), n, ulo,
                          uhi, lo_incl, hi_incl,
                          bcounts.data_ptr<int64_t>(),
                          total.data_ptr<int64_t>(), stream);
  int64_t n_out = total.cpu().item<int64_t>();
  auto out = torch::empty({n_out}, opts);
  if (n_out > 0) {
    hsk::select_range_emit((const uint64_t*)keys.data_ptr<int64_t>(), n, ulo,
                           uhi, lo_incl, hi_incl,
                           bcounts.data_ptr<int64_t>(),
                           out.data_ptr<int64_t>(), stream);
  }
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("murmur3_bucket", &murmur3_bucket, "Spark-compatible murmur3 bucket",
        py::arg("keys"), py::arg("num_buckets"),
        py::arg("masks") = std::vector<torch::Tensor>{});
  m.def("normalize_key", &normalize_key, "order-preserving u64 key");
  m.def("radix_sort_pairs", &radix_sort_pairs, "stable LSD radix sort");
  m.def("merge_join", &merge_join, "segmented sorted merge join");
  m.def("select_range_u64", &select_range_u64_pub, "range filter compaction");
  m.def("isin_sorted", &isin_sorted, "sorted-set membership");
  m.def("segmented_minmax", &segmented_minmax, "per-segment min/max");
  m.def("bloom_build", &bloom_build, "bloom filter build");
  m.def("bloom_probe", &bloom_probe, "bloom filter probe");
  m.def("bloom_probe_many", &bloom_probe_many,
        "K9 device sketch-predicate probe over per-file filters");
  m.def("zorder_key", &zorder_key, "z-order bit interleave");
  m.def("gather_rows", &gather_rows, "row gather by index");
  m.def("copy_unaligned", &copy_unaligned,
        "device parquet page decode (unaligned copy)");
  m.def("run_merge_perm", &run_merge_perm,
        "segmented two-sorted-run merge permutation");
  m.def("snappy_decompress", &snappy_decompress,
        "device snappy raw-block page decompression; returns status");
  m.def("decode_def_levels", &decode_def_levels,
        "host RLE-hybrid def-level decode -> bool validity");
  m.def("parse_rle_runs_batch", &parse_rle_runs_batch,
        py::call_guard<py::gil_scoped_release>(),
        "batched RLE-hybrid run parse over many pages");
  m.def("parse_rle_runs", &parse_rle_runs,
        py::call_guard<py::gil_scoped_release>(),
        "host parse of an RLE/bit-packed hybrid run table");
  m.def("parse_byte_arrays", &parse_byte_arrays,
        py::call_guard<py::gil_scoped_release>(),
        "host PLAIN byte-array page parse -> (offsets, bytes)");
  m.def("host_memcpy", &host_memcpy,
        "GIL-released buffer -> cpu-u8-tensor memcpy");
  m.def("rle_decode", &rle_decode,
        "device RLE/bit-packed dictionary-index decode");
}
