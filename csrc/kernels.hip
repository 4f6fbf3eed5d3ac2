// bodo_amd gfx950 kernels: row hashing, datetime extraction, string gather,
// hash groupby (open addressing + atomics), hash join (bucket chains).
//
// Design notes (MI355X/CDNA4):
//  * memory-bound table kernels: 256-thread blocks (4 waves), grid capped at
//    2048 blocks with grid-stride loops (guide Guideline 11)
//  * hash tables live in HBM (groups/build sides up to hundreds of GB fit in
//    288 GB); LDS pre-aggregation fast path for low-cardinality groupbys is
//    in groupby_lds.hip
//  * all hashes bit-match the host path (ops/__init__.py) so CPU ranks and
//    GPU ranks can share shuffles
//
// Reference behavioral spec (no code reuse): bodo/libs/_array_hash.cpp,
// bodo/libs/streaming/_groupby.cpp, _join.cpp, _datetime_ext.cpp.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include <vector>

#include "common.h"

#define CHECK_HIP(x)                                                    \
  do {                                                                  \
    hipError_t e = (x);                                                 \
    TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));  \
  } while (0)

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// ---------------------------------------------------------------------
// ColumnDesc marshalling
// ---------------------------------------------------------------------

struct DescSet {
  torch::Tensor dev_buf;  // device copy of ColumnDesc[]
  std::vector<ColumnDesc> host;
  ColumnDesc* ptr() { return (ColumnDesc*)dev_buf.data_ptr(); }
};

static ColumnDesc make_desc(const torch::Tensor& data,
                            const c10::optional<torch::Tensor>& mask,
                            const c10::optional<torch::Tensor>& offsets,
                            const c10::optional<torch::Tensor>& aux,
                            int64_t dtype, int64_t n) {
  ColumnDesc d;
  d.data = data.numel() ? data.data_ptr() : nullptr;
  d.offsets = offsets.has_value() ? (const int64_t*)offsets->data_ptr() : nullptr;
  d.mask = mask.has_value() ? (const uint8_t*)mask->data_ptr() : nullptr;
  d.aux = aux.has_value() ? (const uint64_t*)aux->data_ptr() : nullptr;
  d.dtype = (int)dtype;
  d.n = n;
  return d;
}

static DescSet upload_descs(const std::vector<ColumnDesc>& descs,
                            const torch::Device& dev) {
  DescSet s;
  s.host = descs;
  auto cpu = torch::from_blob((void*)s.host.data(),
                              {(int64_t)(descs.size() * sizeof(ColumnDesc))},
                              torch::kUInt8);
  s.dev_buf = cpu.to(dev, /*non_blocking=*/false);
  return s;
}

static DescSet build_descset(
    const std::vector<torch::Tensor>& datas,
    const std::vector<c10::optional<torch::Tensor>>& masks,
    const std::vector<c10::optional<torch::Tensor>>& offsets,
    const std::vector<c10::optional<torch::Tensor>>& auxs,
    const std::vector<int64_t>& dtypes, int64_t n) {
  std::vector<ColumnDesc> descs;
  for (size_t i = 0; i < datas.size(); ++i) {
    descs.push_back(make_desc(datas[i], masks[i], offsets[i], auxs[i],
                              dtypes[i], n));
  }
  return upload_descs(descs, datas[0].device());
}

// ---------------------------------------------------------------------
// row hashing
// ---------------------------------------------------------------------

__global__ void hash_columns_kernel(const ColumnDesc* __restrict__ cols,
                                    int ncols, uint64_t seed,
                                    uint64_t* __restrict__ out, int64_t n) {
  GRID_STRIDE_LOOP(i, n) {
    uint64_t acc = 0;
    for (int c = 0; c < ncols; ++c) {
      uint64_t h = is_valid_at(cols[c], i) ? hash_value(cols[c], i) : NULL_HASH;
      if (seed) h = mix64(h ^ seed);
      acc = (c == 0) ? h : mix64(acc * GOLDEN + h);
    }
    out[i] = acc;
  }
}

torch::Tensor hash_columns(
    std::vector<torch::Tensor> datas,
    std::vector<c10::optional<torch::Tensor>> masks,
    std::vector<c10::optional<torch::Tensor>> offsets,
    std::vector<c10::optional<torch::Tensor>> auxs,
    std::vector<int64_t> dtypes, int64_t n, int64_t seed) {
  auto dev = datas[0].device();
  auto ds = build_descset(datas, masks, offsets, auxs, dtypes, n);
  auto out = torch::empty({n}, torch::dtype(torch::kInt64).device(dev));
  int block = 256;
  hipLaunchKernelGGL(hash_columns_kernel, dim3(grid_for(n, block)),
                     dim3(block), 0, cur_stream(), ds.ptr(),
                     (int)datas.size(), (uint64_t)seed,
                     (uint64_t*)out.data_ptr(), n);
  CHECK_HIP(hipGetLastError());
  return out;
}

// ---------------------------------------------------------------------
// datetime extraction (Howard Hinnant civil-from-days algorithm)
// ---------------------------------------------------------------------

DEV_INLINE int64_t floordiv(int64_t a, int64_t b) {
  int64_t q = a / b;
  return (a % b != 0 && ((a < 0) != (b < 0))) ? q - 1 : q;
}

DEV_INLINE void civil_from_days(int64_t z, int& y, unsigned& m, unsigned& d) {
  z += 719468;
  const int64_t era = (z >= 0 ? z : z - 146096) / 146097;
  const unsigned doe = (unsigned)(z - era * 146097);
  const unsigned yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  const int64_t y_ = (int64_t)yoe + era * 400;
  const unsigned doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  const unsigned mp = (5 * doy + 2) / 153;
  d = doy - (153 * mp + 2) / 5 + 1;
  m = mp < 10 ? mp + 3 : mp - 9;
  y = (int)(y_ + (m <= 2));
}

enum DtField : int {
  DT_YEAR = 0, DT_MONTH = 1, DT_DAY = 2, DT_HOUR = 3, DT_MINUTE = 4,
  DT_SECOND = 5, DT_DAYOFWEEK = 6, DT_DAYOFYEAR = 7, DT_QUARTER = 8,
  DT_DATE = 9, DT_NORMALIZE = 10,
};

#define NS_PER_DAY 86400000000000LL

template <typename OUT>
__global__ void dt_field_kernel(const int64_t* __restrict__ ts, int is_date32,
                                const int32_t* __restrict__ ts32, int field,
                                OUT* __restrict__ out, int64_t n) {
  GRID_STRIDE_LOOP(i, n) {
    int64_t days, ns_in_day;
    if (is_date32) {
      days = ts32[i];
      ns_in_day = 0;
    } else {
      int64_t v = ts[i];
      days = floordiv(v, NS_PER_DAY);
      ns_in_day = v - days * NS_PER_DAY;
    }
    OUT r = 0;
    switch (field) {
      case DT_DATE: r = (OUT)days; break;
      case DT_NORMALIZE: r = (OUT)(days * NS_PER_DAY); break;
      case DT_HOUR: r = (OUT)(ns_in_day / 3600000000000LL); break;
      case DT_MINUTE: r = (OUT)((ns_in_day / 60000000000LL) % 60); break;
      case DT_SECOND: r = (OUT)((ns_in_day / 1000000000LL) % 60); break;
      case DT_DAYOFWEEK: {
        // 1970-01-01 is a Thursday = 3 (Mon=0)
        int64_t dow = (days + 3) % 7;
        if (dow < 0) dow += 7;
        r = (OUT)dow;
        break;
      }
      default: {
        int y; unsigned m, d;
        civil_from_days(days, y, m, d);
        if (field == DT_YEAR) r = (OUT)y;
        else if (field == DT_MONTH) r = (OUT)m;
        else if (field == DT_DAY) r = (OUT)d;
        else if (field == DT_QUARTER) r = (OUT)((m - 1) / 3 + 1);
        else if (field == DT_DAYOFYEAR) {
          // days since Jan 1 of year y
          int64_t jan1 = days;
          // compute days-from-civil(y,1,1)
          int64_t yy = y;
          yy -= 1 <= 2;
          const int64_t era = (yy >= 0 ? yy : yy - 399) / 400;
          const unsigned yoe = (unsigned)(yy - era * 400);
          const unsigned doy0 = (153 * (1 + 9) + 2) / 5 + 1 - 1;
          const unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy0;
          jan1 = era * 146097 + (int64_t)doe - 719468;
          r = (OUT)(days - jan1 + 1);
        }
      }
    }
    out[i] = r;
  }
}

torch::Tensor dt_field(torch::Tensor data, int64_t is_date32, int64_t field,
                       int64_t out_kind) {
  int64_t n = data.numel();
  auto dev = data.device();
  torch::Tensor out;
  int block = 256;
  auto grid = dim3(grid_for(n, block));
  const int64_t* ts = is_date32 ? nullptr : (const int64_t*)data.data_ptr();
  const int32_t* ts32 = is_date32 ? (const int32_t*)data.data_ptr() : nullptr;
  if (out_kind == 0) {  // int16
    out = torch::empty({n}, torch::dtype(torch::kInt16).device(dev));
    hipLaunchKernelGGL(dt_field_kernel<int16_t>, grid, dim3(block), 0,
                       cur_stream(), ts, (int)is_date32, ts32, (int)field,
                       (int16_t*)out.data_ptr(), n);
  } else if (out_kind == 1) {  // int32
    out = torch::empty({n}, torch::dtype(torch::kInt32).device(dev));
    hipLaunchKernelGGL(dt_field_kernel<int32_t>, grid, dim3(block), 0,
                       cur_stream(), ts, (int)is_date32, ts32, (int)field,
                       (int32_t*)out.data_ptr(), n);
  } else {  // int64
    out = torch::empty({n}, torch::dtype(torch::kInt64).device(dev));
    hipLaunchKernelGGL(dt_field_kernel<int64_t>, grid, dim3(block), 0,
                       cur_stream(), ts, (int)is_date32, ts32, (int)field,
                       (int64_t*)out.data_ptr(), n);
  }
  CHECK_HIP(hipGetLastError());
  return out;
}

// ---------------------------------------------------------------------
// string gather: one wave per row, 8-byte chunks
// ---------------------------------------------------------------------

__global__ void string_gather_kernel(
    const uint8_t* __restrict__ src, const int64_t* __restrict__ src_off,
    const int64_t* __restrict__ idx, const int64_t* __restrict__ dst_off,
    uint8_t* __restrict__ dst, int64_t n_rows) {
  // one wave (64 lanes) per row; lanes copy bytes cooperatively
  int64_t row = blockIdx.x * (int64_t)(blockDim.x / WAVE) + threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  int64_t stride = (int64_t)gridDim.x * (blockDim.x / WAVE);
  for (; row < n_rows; row += stride) {
    int64_t s = src_off[idx[row]];
    int64_t len = src_off[idx[row] + 1] - s;
    int64_t d = dst_off[row];
    for (int64_t k = lane; k < len; k += WAVE) {
      dst[d + k] = src[s + k];
    }
  }
}

std::vector<torch::Tensor> gather_string(torch::Tensor bytes,
                                         torch::Tensor offsets,
                                         torch::Tensor idx) {
  int64_t n = idx.numel();
  auto dev = bytes.device();
  auto starts = offsets.index({idx});
  auto lens = offsets.index({idx + 1}) - starts;
  auto new_off = torch::zeros({n + 1}, torch::dtype(torch::kInt64).device(dev));
  auto off_tail = new_off.slice(0, 1, n + 1);
  at::cumsum_out(off_tail, lens, 0);
  int64_t total = n ? new_off[n].item<int64_t>() : 0;
  auto out = torch::empty({total}, torch::dtype(torch::kUInt8).device(dev));
  if (n && total) {
    int block = 256;
    int waves_per_block = block / WAVE;
    int grid = (int)std::min<int64_t>((n + waves_per_block - 1) / waves_per_block, 2048);
    hipLaunchKernelGGL(string_gather_kernel, dim3(grid), dim3(block), 0,
                       cur_stream(), (const uint8_t*)bytes.data_ptr(),
                       (const int64_t*)offsets.data_ptr(),
                       (const int64_t*)idx.data_ptr(),
                       (const int64_t*)new_off.data_ptr(),
                       (uint8_t*)out.data_ptr(), n);
    CHECK_HIP(hipGetLastError());
  }
  return {out, new_off};
}

// ---------------------------------------------------------------------
// hash groupby: open-addressing table in HBM (reference semantics:
// streaming/_groupby.cpp HashGroupbyTable)
// ---------------------------------------------------------------------

__global__ void gb_insert_kernel(const ColumnDesc* __restrict__ cols, int ncols,
                                 const uint64_t* __restrict__ hashes,
                                 uint32_t* __restrict__ slots, uint64_t cap_mask,
                                 uint32_t* __restrict__ row_slot, int64_t n,
                                 int64_t row0, const uint64_t* __restrict__ all_hashes) {
  // rows [row0, row0+n) of the table; hashes/row_slot pre-offset by caller,
  // all_hashes/cols indexed globally (slots store global row+1)
  GRID_STRIDE_LOOP(i, n) {
    uint64_t h = hashes[i];
    int64_t gi = row0 + i;
    uint64_t s = h & cap_mask;
    while (true) {
      uint32_t old = atomicCAS(&slots[s], 0u, (uint32_t)(gi + 1));
      if (old == 0u) {  // we claimed the slot: new group
        row_slot[i] = (uint32_t)s;
        break;
      }
      int64_t cand = (int64_t)old - 1;
      if (all_hashes[cand] == h && rows_eq(cols, ncols, cand, gi)) {
        row_slot[i] = (uint32_t)s;
        break;
      }
      s = (s + 1) & cap_mask;
    }
  }
}

// two-pass group-id assignment: pass 1 counts occupied slots per block
// (no atomics), host cumsums 2048 values, pass 2 assigns gids via an LDS
// counter (one global atomic total: zero).  Replaces a single-global-counter
// atomicAdd that measured 43% of NYC-taxi Q1 step time at 300M rows.
__global__ void gb_count_kernel(const uint32_t* __restrict__ slots,
                                int64_t chunk, int64_t cap,
                                int64_t* __restrict__ block_counts) {
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t stop = min(start + chunk, cap);
  int64_t local = 0;
  for (int64_t s = start + threadIdx.x; s < stop; s += blockDim.x) {
    local += (slots[s] != 0u);
  }
  __shared__ int64_t red[256];
  red[threadIdx.x] = local;
  __syncthreads();
  for (int w = blockDim.x / 2; w > 0; w >>= 1) {
    if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) block_counts[blockIdx.x] = red[0];
}

__global__ void gb_rowgid_kernel(const uint32_t* __restrict__ row_slot,
                                 const int32_t* __restrict__ slot_gid,
                                 int32_t* __restrict__ row_gid, int64_t n);

// Packed-key fast path: keys pre-packed into one u64 (< 2^63), slots hold
// the KEY itself (CAS, sentinel ~0) so probing never dereferences candidate
// rows or a hash array; the slot hash is computed in-kernel.
#define PACKED_EMPTY 0xFFFFFFFFFFFFFFFFULL

__global__ void gb_insert_packed_kernel(const int64_t* __restrict__ keys,
                                        unsigned long long* __restrict__ slot_keys,
                                        uint32_t* __restrict__ slot_rows,
                                        uint64_t cap_mask,
                                        uint32_t* __restrict__ row_slot,
                                        int64_t n) {
  GRID_STRIDE_LOOP(i, n) {
    unsigned long long key = (unsigned long long)keys[i];
    uint64_t s = mix64(key) & cap_mask;
    while (true) {
      unsigned long long old = slot_keys[s];
      if (old == key) {
        row_slot[i] = (uint32_t)s;
        break;
      }
      if (old == PACKED_EMPTY) {
        old = atomicCAS(&slot_keys[s], PACKED_EMPTY, key);
        if (old == PACKED_EMPTY) {
          slot_rows[s] = (uint32_t)(i + 1);
          row_slot[i] = (uint32_t)s;
          break;
        }
        if (old == key) {
          row_slot[i] = (uint32_t)s;
          break;
        }
      }
      s = (s + 1) & cap_mask;
    }
  }
}

__global__ void gb_assign_gid_packed_kernel(
    const unsigned long long* __restrict__ slot_keys,
    const uint32_t* __restrict__ slot_rows, int64_t chunk, int64_t cap,
    const int64_t* __restrict__ block_base, int32_t* __restrict__ slot_gid,
    int64_t* __restrict__ uniq_rows) {
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t stop = min(start + chunk, cap);
  __shared__ int lds_cnt;
  if (threadIdx.x == 0) lds_cnt = 0;
  __syncthreads();
  int64_t base = block_base[blockIdx.x];
  for (int64_t s = start + threadIdx.x; s < stop; s += blockDim.x) {
    if (slot_keys[s] != PACKED_EMPTY) {
      int32_t gid = (int32_t)(base + atomicAdd(&lds_cnt, 1));
      slot_gid[s] = gid;
      uniq_rows[gid] = (int64_t)slot_rows[s] - 1;
    }
  }
}

__global__ void gb_count_packed_kernel(
    const unsigned long long* __restrict__ slot_keys, int64_t chunk,
    int64_t cap, int64_t* __restrict__ block_counts) {
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t stop = min(start + chunk, cap);
  int64_t local = 0;
  for (int64_t s = start + threadIdx.x; s < stop; s += blockDim.x) {
    local += (slot_keys[s] != PACKED_EMPTY);
  }
  __shared__ int64_t red[256];
  red[threadIdx.x] = local;
  __syncthreads();
  for (int w = blockDim.x / 2; w > 0; w >>= 1) {
    if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) block_counts[blockIdx.x] = red[0];
}

std::vector<torch::Tensor> groupby_build_packed(torch::Tensor keys) {
  auto dev = keys.device();
  int64_t n = keys.numel();
  int block = 256;
  // 2^25 slots x 8 B = 256 MB: Infinity-Cache resident; grow on pressure
  int64_t cap = 16;
  while (cap < 2 * n) cap <<= 1;
  if (cap > (1 << 25)) cap = 1 << 25;
  auto row_slot = torch::empty({n}, torch::dtype(torch::kInt32).device(dev));
  torch::Tensor slot_keys, slot_rows;
  const int64_t chunk_rows = 1 << 27;
  while (true) {
    slot_keys = torch::full({cap}, -1, torch::dtype(torch::kInt64).device(dev));
    slot_rows = torch::empty({cap}, torch::dtype(torch::kInt32).device(dev));
    bool grown = false;
    for (int64_t start = 0; start < n; start += chunk_rows) {
      int64_t cnt = std::min(chunk_rows, n - start);
      hipLaunchKernelGGL(gb_insert_packed_kernel, dim3(grid_for(cnt, block)),
                         dim3(block), 0, cur_stream(),
                         (const int64_t*)keys.data_ptr() + start,
                         (unsigned long long*)slot_keys.data_ptr(),
                         (uint32_t*)slot_rows.data_ptr(), (uint64_t)(cap - 1),
                         (uint32_t*)row_slot.data_ptr() + start, cnt);
      CHECK_HIP(hipGetLastError());
      if (start + cnt < n && cap < 2 * n) {
        int64_t occupied = (slot_keys != -1).sum().item<int64_t>();
        if (occupied * 2 > cap) {
          int64_t want = cap * 8;
          while (want < 2 * std::min(n, occupied * 8)) want <<= 1;
          if (want > 2 * n) {
            int64_t c2 = 16;
            while (c2 < 2 * n) c2 <<= 1;
            want = c2;
          }
          cap = want;
          grown = true;
          break;
        }
      }
    }
    if (!grown) break;
  }
  int nblocks = 2048;
  int64_t chunk = (cap + nblocks - 1) / nblocks;
  auto block_counts = torch::zeros({nblocks},
                                   torch::dtype(torch::kInt64).device(dev));
  hipLaunchKernelGGL(gb_count_packed_kernel, dim3(nblocks), dim3(block), 0,
                     cur_stream(),
                     (const unsigned long long*)slot_keys.data_ptr(), chunk,
                     cap, (int64_t*)block_counts.data_ptr());
  CHECK_HIP(hipGetLastError());
  auto block_base = torch::zeros({nblocks}, torch::dtype(torch::kInt64).device(dev));
  auto bb_tail = block_base.slice(0, 1, nblocks);
  at::cumsum_out(bb_tail, block_counts.slice(0, 0, nblocks - 1), 0);
  int64_t ngroups = block_counts.sum().item<int64_t>();
  auto slot_gid = torch::empty({cap}, torch::dtype(torch::kInt32).device(dev));
  auto uniq_rows = torch::empty({std::max<int64_t>(ngroups, 1)},
                                torch::dtype(torch::kInt64).device(dev));
  hipLaunchKernelGGL(gb_assign_gid_packed_kernel, dim3(nblocks), dim3(block),
                     0, cur_stream(),
                     (const unsigned long long*)slot_keys.data_ptr(),
                     (const uint32_t*)slot_rows.data_ptr(), chunk, cap,
                     (const int64_t*)block_base.data_ptr(),
                     (int32_t*)slot_gid.data_ptr(),
                     (int64_t*)uniq_rows.data_ptr());
  CHECK_HIP(hipGetLastError());
  auto row_gid = torch::empty({n}, torch::dtype(torch::kInt32).device(dev));
  hipLaunchKernelGGL(gb_rowgid_kernel, dim3(grid_for(n, block)), dim3(block),
                     0, cur_stream(), (const uint32_t*)row_slot.data_ptr(),
                     (const int32_t*)slot_gid.data_ptr(),
                     (int32_t*)row_gid.data_ptr(), n);
  CHECK_HIP(hipGetLastError());
  return {row_gid, uniq_rows.slice(0, 0, std::max<int64_t>(ngroups, 0))};
}

__global__ void gb_assign_gid_kernel(const uint32_t* __restrict__ slots,
                                     int64_t chunk, int64_t cap,
                                     const int64_t* __restrict__ block_base,
                                     int32_t* __restrict__ slot_gid,
                                     int64_t* __restrict__ uniq_rows) {
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t stop = min(start + chunk, cap);
  __shared__ int lds_cnt;
  if (threadIdx.x == 0) lds_cnt = 0;
  __syncthreads();
  int64_t base = block_base[blockIdx.x];
  for (int64_t s = start + threadIdx.x; s < stop; s += blockDim.x) {
    uint32_t v = slots[s];
    if (v != 0u) {
      int32_t gid = (int32_t)(base + atomicAdd(&lds_cnt, 1));
      slot_gid[s] = gid;
      uniq_rows[gid] = (int64_t)v - 1;
    }
  }
}

__global__ void gb_rowgid_kernel(const uint32_t* __restrict__ row_slot,
                                 const int32_t* __restrict__ slot_gid,
                                 int32_t* __restrict__ row_gid, int64_t n) {
  GRID_STRIDE_LOOP(i, n) { row_gid[i] = slot_gid[row_slot[i]]; }
}

std::vector<torch::Tensor> groupby_build(
    std::vector<torch::Tensor> datas,
    std::vector<c10::optional<torch::Tensor>> masks,
    std::vector<c10::optional<torch::Tensor>> offsets,
    std::vector<c10::optional<torch::Tensor>> auxs,
    std::vector<int64_t> dtypes, int64_t n, torch::Tensor hashes) {
  auto dev = datas[0].device();
  auto ds = build_descset(datas, masks, offsets, auxs, dtypes, n);
  // Start with an Infinity-Cache-resident table (2^26 slots = 256 MB):
  // the random atomicCAS probes then hit L3 instead of HBM (guide §2).
  // Insert in chunks, checking the load factor between chunks; grow 8x and
  // re-insert when it crosses 1/2 (few groupbys exceed 32M groups/rank).
  int64_t cap = 16;
  while (cap < 2 * n) cap <<= 1;
  if (cap > (1 << 26)) cap = 1 << 26;
  int block = 256;
  auto row_slot = torch::empty({n}, torch::dtype(torch::kInt32).device(dev));
  torch::Tensor slots;
  const int64_t chunk_rows = 1 << 27;
  while (true) {
    slots = torch::zeros({cap}, torch::dtype(torch::kInt32).device(dev));
    bool grown = false;
    for (int64_t start = 0; start < n; start += chunk_rows) {
      int64_t cnt = std::min(chunk_rows, n - start);
      hipLaunchKernelGGL(gb_insert_kernel, dim3(grid_for(cnt, block)),
                         dim3(block), 0, cur_stream(), ds.ptr(),
                         (int)datas.size(),
                         (const uint64_t*)hashes.data_ptr() + start,
                         (uint32_t*)slots.data_ptr(), (uint64_t)(cap - 1),
                         (uint32_t*)row_slot.data_ptr() + start, cnt, start,
                         (const uint64_t*)hashes.data_ptr());
      CHECK_HIP(hipGetLastError());
      if (start + cnt < n && cap < 2 * n) {
        int64_t occupied = (slots != 0).sum().item<int64_t>();
        if (occupied * 2 > cap) {
          int64_t want = cap * 8;
          while (want < 2 * std::min(n, occupied * 8)) want <<= 1;
          if (want > 2 * n) {
            int64_t c2 = 16;
            while (c2 < 2 * n) c2 <<= 1;
            want = c2;
          }
          cap = want;
          grown = true;
          break;  // rebuild from scratch at the bigger capacity
        }
      }
    }
    if (!grown) break;
  }
  auto slot_gid = torch::empty({cap}, torch::dtype(torch::kInt32).device(dev));
  auto uniq_rows = torch::empty({n > 0 ? n : 1},
                                torch::dtype(torch::kInt64).device(dev));
  int nblocks = 2048;
  int64_t chunk = (cap + nblocks - 1) / nblocks;
  auto block_counts = torch::zeros({nblocks},
                                   torch::dtype(torch::kInt64).device(dev));
  hipLaunchKernelGGL(gb_count_kernel, dim3(nblocks), dim3(block), 0,
                     cur_stream(), (const uint32_t*)slots.data_ptr(), chunk,
                     cap, (int64_t*)block_counts.data_ptr());
  CHECK_HIP(hipGetLastError());
  auto block_base = torch::zeros({nblocks}, torch::dtype(torch::kInt64).device(dev));
  auto bb_tail = block_base.slice(0, 1, nblocks);
  at::cumsum_out(bb_tail, block_counts.slice(0, 0, nblocks - 1), 0);
  int64_t ngroups = block_counts.sum().item<int64_t>();
  hipLaunchKernelGGL(gb_assign_gid_kernel, dim3(nblocks), dim3(block), 0,
                     cur_stream(), (const uint32_t*)slots.data_ptr(), chunk,
                     cap, (const int64_t*)block_base.data_ptr(),
                     (int32_t*)slot_gid.data_ptr(),
                     (int64_t*)uniq_rows.data_ptr());
  CHECK_HIP(hipGetLastError());
  auto row_gid = torch::empty({n}, torch::dtype(torch::kInt32).device(dev));
  hipLaunchKernelGGL(gb_rowgid_kernel, dim3(grid_for(n, block)), dim3(block),
                     0, cur_stream(), (const uint32_t*)row_slot.data_ptr(),
                     (const int32_t*)slot_gid.data_ptr(),
                     (int32_t*)row_gid.data_ptr(), n);
  CHECK_HIP(hipGetLastError());
  return {row_gid, uniq_rows.slice(0, 0, ngroups)};
}

// ------------------------------------------------------------ agg update

enum AggOp : int {
  AGG_SUM_F64 = 0, AGG_SUM_I64 = 1, AGG_COUNT = 2, AGG_MIN_F64 = 3,
  AGG_MAX_F64 = 4, AGG_MIN_I64 = 5, AGG_MAX_I64 = 6, AGG_SIZE = 7,
  AGG_FIRST_ROW = 8, AGG_LAST_ROW = 9, AGG_PROD_F64 = 10,
};

template <typename T>
DEV_INLINE double load_as_f64(const T* p, int64_t i) { return (double)p[i]; }

// value loader switch: returns value as double or int64 depending on op
__global__ void agg_update_kernel(const ColumnDesc col,
                                  const int32_t* __restrict__ row_gid,
                                  int op, void* __restrict__ acc,
                                  int64_t* __restrict__ cnt, int64_t n) {
  GRID_STRIDE_LOOP(i, n) {
    int32_t g = row_gid[i];
    if (op == AGG_SIZE) {
      atomicAdd((unsigned long long*)&cnt[g], 1ull);
      continue;
    }
    bool valid = is_valid_at(col, i);
    double dv = 0.0;
    int64_t iv = 0;
    if (valid) {
      switch (col.dtype) {
        case BT_INT8: iv = ((const int8_t*)col.data)[i]; dv = (double)iv; break;
        case BT_UINT8: case BT_BOOL: iv = ((const uint8_t*)col.data)[i]; dv = (double)iv; break;
        case BT_INT16: case BT_UINT16: iv = ((const int16_t*)col.data)[i]; dv = (double)iv; break;
        case BT_INT32: case BT_UINT32: case BT_DATE32: case BT_DICT:
          iv = ((const int32_t*)col.data)[i]; dv = (double)iv; break;
        case BT_INT64: case BT_UINT64: case BT_TIMESTAMP_NS:
          iv = ((const int64_t*)col.data)[i]; dv = (double)iv; break;
        case BT_FLOAT32: {
          float f = ((const float*)col.data)[i];
          valid = !(f != f);
          dv = (double)f; iv = (int64_t)f;
          break;
        }
        case BT_FLOAT64: {
          double f = ((const double*)col.data)[i];
          valid = !(f != f);
          dv = f; iv = (int64_t)f;
          break;
        }
      }
    }
    if (!valid) continue;
    switch (op) {
      case AGG_SUM_F64: atomicAdd((double*)acc + g, dv); break;
      case AGG_SUM_I64: atomicAdd((unsigned long long*)acc + g, (unsigned long long)iv); break;
      case AGG_COUNT: break;  // counted below
      case AGG_MIN_F64: atomic_min_f64((double*)acc + g, dv); break;
      case AGG_MAX_F64: atomic_max_f64((double*)acc + g, dv); break;
      case AGG_MIN_I64: atomic_min_i64((int64_t*)acc + g, iv); break;
      case AGG_MAX_I64: atomic_max_i64((int64_t*)acc + g, iv); break;
      case AGG_FIRST_ROW: atomic_min_i64((int64_t*)acc + g, (int64_t)i); break;
      case AGG_LAST_ROW: atomic_max_i64((int64_t*)acc + g, (int64_t)i); break;
      case AGG_PROD_F64: {
        // CAS-loop multiply
        unsigned long long* a = (unsigned long long*)acc + g;
        unsigned long long old = *a, assumed;
        do {
          assumed = old;
          double nv = __longlong_as_double(assumed) * dv;
          old = atomicCAS(a, assumed, (unsigned long long)__double_as_longlong(nv));
        } while (old != assumed);
        break;
      }
    }
    if (cnt != nullptr) atomicAdd((unsigned long long*)&cnt[g], 1ull);
  }
}

// LDS pre-aggregation path for low-cardinality groupbys (TPC-H Q1 shape):
// per-block accumulators in LDS, one global atomic per (block, group).
// Replaces 60M global atomics on 6 slots (measured 243 ms/call) with
// 60M LDS atomics + ~2048*6 global ones.
#define LDS_AGG_MAX_GROUPS 2048

__global__ void agg_update_lds_kernel(const ColumnDesc col,
                                      const int32_t* __restrict__ row_gid,
                                      int op, int ngroups,
                                      void* __restrict__ acc,
                                      int64_t* __restrict__ cnt, int64_t n,
                                      double init_f, int64_t init_i) {
  __shared__ double lacc[LDS_AGG_MAX_GROUPS];
  __shared__ long long lcnt[LDS_AGG_MAX_GROUPS];
  bool f64_acc = (op == AGG_SUM_F64 || op == AGG_MIN_F64 || op == AGG_MAX_F64);
  for (int g = threadIdx.x; g < ngroups; g += blockDim.x) {
    lacc[g] = f64_acc ? init_f : __longlong_as_double(init_i);
    lcnt[g] = 0;
  }
  __syncthreads();
  GRID_STRIDE_LOOP(i, n) {
    int32_t g = row_gid[i];
    if (op == AGG_SIZE) {
      atomicAdd((unsigned long long*)&lcnt[g], 1ull);
      continue;
    }
    bool valid = is_valid_at(col, i);
    double dv = 0.0;
    int64_t iv = 0;
    if (valid) {
      switch (col.dtype) {
        case BT_INT8: iv = ((const int8_t*)col.data)[i]; dv = (double)iv; break;
        case BT_UINT8: case BT_BOOL: iv = ((const uint8_t*)col.data)[i]; dv = (double)iv; break;
        case BT_INT16: case BT_UINT16: iv = ((const int16_t*)col.data)[i]; dv = (double)iv; break;
        case BT_INT32: case BT_UINT32: case BT_DATE32: case BT_DICT:
          iv = ((const int32_t*)col.data)[i]; dv = (double)iv; break;
        case BT_INT64: case BT_UINT64: case BT_TIMESTAMP_NS:
          iv = ((const int64_t*)col.data)[i]; dv = (double)iv; break;
        case BT_FLOAT32: { float f = ((const float*)col.data)[i]; valid = !(f != f); dv = (double)f; iv = (int64_t)f; break; }
        case BT_FLOAT64: { double f = ((const double*)col.data)[i]; valid = !(f != f); dv = f; iv = (int64_t)f; break; }
      }
    }
    if (!valid) continue;
    switch (op) {
      case AGG_SUM_F64: atomicAdd(&lacc[g], dv); break;
      case AGG_SUM_I64: atomicAdd((unsigned long long*)&lacc[g], (unsigned long long)iv); break;
      case AGG_COUNT: break;
      case AGG_MIN_F64: atomic_min_f64(&lacc[g], dv); break;
      case AGG_MAX_F64: atomic_max_f64(&lacc[g], dv); break;
      case AGG_MIN_I64: atomic_min_i64((int64_t*)&lacc[g], iv); break;
      case AGG_MAX_I64: atomic_max_i64((int64_t*)&lacc[g], iv); break;
      case AGG_FIRST_ROW: atomic_min_i64((int64_t*)&lacc[g], (int64_t)i); break;
      case AGG_LAST_ROW: atomic_max_i64((int64_t*)&lacc[g], (int64_t)i); break;
    }
    atomicAdd((unsigned long long*)&lcnt[g], 1ull);
  }
  __syncthreads();
  // merge block partials into global accumulators
  for (int g = threadIdx.x; g < ngroups; g += blockDim.x) {
    long long c = lcnt[g];
    if (cnt != nullptr && c) atomicAdd((unsigned long long*)&cnt[g], (unsigned long long)c);
    if (op == AGG_SIZE || op == AGG_COUNT) continue;
    if (c == 0) continue;
    switch (op) {
      case AGG_SUM_F64: atomicAdd((double*)acc + g, lacc[g]); break;
      case AGG_SUM_I64: atomicAdd((unsigned long long*)acc + g,
                                  (unsigned long long)__double_as_longlong(lacc[g])); break;
      case AGG_MIN_F64: atomic_min_f64((double*)acc + g, lacc[g]); break;
      case AGG_MAX_F64: atomic_max_f64((double*)acc + g, lacc[g]); break;
      case AGG_MIN_I64: atomic_min_i64((int64_t*)acc + g, __double_as_longlong(lacc[g])); break;
      case AGG_MAX_I64: atomic_max_i64((int64_t*)acc + g, __double_as_longlong(lacc[g])); break;
      case AGG_FIRST_ROW: atomic_min_i64((int64_t*)acc + g, __double_as_longlong(lacc[g])); break;
      case AGG_LAST_ROW: atomic_max_i64((int64_t*)acc + g, __double_as_longlong(lacc[g])); break;
    }
  }
}

// fused multi-aggregate: one pass over row_gid updating up to 4 aggregates
// (taxi Q1: count + sum + count in ONE read of the gid array)
#define MAX_FUSED_AGGS 4

struct FusedAgg {
  ColumnDesc col;
  int op;
  void* acc;
  int64_t* cnt;
  double init_f;
  int64_t init_i;
};

__global__ void agg_update_fused_kernel(FusedAgg a0, FusedAgg a1, FusedAgg a2,
                                        FusedAgg a3, int n_aggs,
                                        const int32_t* __restrict__ row_gid,
                                        int64_t n) {
  FusedAgg aggs[MAX_FUSED_AGGS] = {a0, a1, a2, a3};
  GRID_STRIDE_LOOP(i, n) {
    int32_t g = row_gid[i];
    for (int k = 0; k < n_aggs; ++k) {
      const ColumnDesc& col = aggs[k].col;
      int op = aggs[k].op;
      void* acc = aggs[k].acc;
      int64_t* cnt = aggs[k].cnt;
      if (op == AGG_SIZE) {
        atomicAdd((unsigned long long*)&cnt[g], 1ull);
        continue;
      }
      bool valid = is_valid_at(col, i);
      double dv = 0.0;
      int64_t iv = 0;
      if (valid) {
        switch (col.dtype) {
          case BT_INT8: iv = ((const int8_t*)col.data)[i]; dv = (double)iv; break;
          case BT_UINT8: case BT_BOOL: iv = ((const uint8_t*)col.data)[i]; dv = (double)iv; break;
          case BT_INT16: case BT_UINT16: iv = ((const int16_t*)col.data)[i]; dv = (double)iv; break;
          case BT_INT32: case BT_UINT32: case BT_DATE32: case BT_DICT:
            iv = ((const int32_t*)col.data)[i]; dv = (double)iv; break;
          case BT_INT64: case BT_UINT64: case BT_TIMESTAMP_NS:
            iv = ((const int64_t*)col.data)[i]; dv = (double)iv; break;
          case BT_FLOAT32: { float f = ((const float*)col.data)[i]; valid = !(f != f); dv = (double)f; iv = (int64_t)f; break; }
          case BT_FLOAT64: { double f = ((const double*)col.data)[i]; valid = !(f != f); dv = f; iv = (int64_t)f; break; }
        }
      }
      if (!valid) continue;
      switch (op) {
        case AGG_SUM_F64: atomicAdd((double*)acc + g, dv); break;
        case AGG_SUM_I64: atomicAdd((unsigned long long*)acc + g, (unsigned long long)iv); break;
        case AGG_COUNT: break;
        case AGG_MIN_F64: atomic_min_f64((double*)acc + g, dv); break;
        case AGG_MAX_F64: atomic_max_f64((double*)acc + g, dv); break;
        case AGG_MIN_I64: atomic_min_i64((int64_t*)acc + g, iv); break;
        case AGG_MAX_I64: atomic_max_i64((int64_t*)acc + g, iv); break;
      }
      if (cnt != nullptr) atomicAdd((unsigned long long*)&cnt[g], 1ull);
    }
  }
}

// LDS pre-aggregation variant of the fused kernel for low-cardinality keys
// (TPC-H Q1: 6 groups, 8 aggregates).  Without it 60M rows x 4 aggs of
// global atomics on 6 slots measured 274 ms/launch; with block-local LDS
// accumulators the global traffic is ~grid*ngroups atomics.
__global__ void agg_update_fused_lds_kernel(
    FusedAgg a0, FusedAgg a1, FusedAgg a2, FusedAgg a3, int n_aggs,
    int ngroups, const int32_t* __restrict__ row_gid, int64_t n) {
  FusedAgg aggs[MAX_FUSED_AGGS] = {a0, a1, a2, a3};
  extern __shared__ double smem[];
  double* lacc = smem;                              // n_aggs * ngroups
  long long* lcnt = (long long*)(smem + (size_t)n_aggs * ngroups);
  for (int j = threadIdx.x; j < n_aggs * ngroups; j += blockDim.x) {
    int k = j / ngroups;
    int op = aggs[k].op;
    bool f64_acc = (op == AGG_SUM_F64 || op == AGG_MIN_F64 || op == AGG_MAX_F64);
    lacc[j] = f64_acc ? aggs[k].init_f : __longlong_as_double(aggs[k].init_i);
    lcnt[j] = 0;
  }
  __syncthreads();
  GRID_STRIDE_LOOP(i, n) {
    int32_t g = row_gid[i];
    for (int k = 0; k < n_aggs; ++k) {
      double* la = lacc + (size_t)k * ngroups;
      long long* lc = lcnt + (size_t)k * ngroups;
      int op = aggs[k].op;
      if (op == AGG_SIZE) {
        atomicAdd((unsigned long long*)&lc[g], 1ull);
        continue;
      }
      const ColumnDesc& col = aggs[k].col;
      bool valid = is_valid_at(col, i);
      double dv = 0.0;
      int64_t iv = 0;
      if (valid) {
        switch (col.dtype) {
          case BT_INT8: iv = ((const int8_t*)col.data)[i]; dv = (double)iv; break;
          case BT_UINT8: case BT_BOOL: iv = ((const uint8_t*)col.data)[i]; dv = (double)iv; break;
          case BT_INT16: case BT_UINT16: iv = ((const int16_t*)col.data)[i]; dv = (double)iv; break;
          case BT_INT32: case BT_UINT32: case BT_DATE32: case BT_DICT:
            iv = ((const int32_t*)col.data)[i]; dv = (double)iv; break;
          case BT_INT64: case BT_UINT64: case BT_TIMESTAMP_NS:
            iv = ((const int64_t*)col.data)[i]; dv = (double)iv; break;
          case BT_FLOAT32: { float f = ((const float*)col.data)[i]; valid = !(f != f); dv = (double)f; iv = (int64_t)f; break; }
          case BT_FLOAT64: { double f = ((const double*)col.data)[i]; valid = !(f != f); dv = f; iv = (int64_t)f; break; }
        }
      }
      if (!valid) continue;
      switch (op) {
        case AGG_SUM_F64: atomicAdd(&la[g], dv); break;
        case AGG_SUM_I64: atomicAdd((unsigned long long*)&la[g], (unsigned long long)iv); break;
        case AGG_COUNT: break;
        case AGG_MIN_F64: atomic_min_f64(&la[g], dv); break;
        case AGG_MAX_F64: atomic_max_f64(&la[g], dv); break;
        case AGG_MIN_I64: atomic_min_i64((int64_t*)&la[g], iv); break;
        case AGG_MAX_I64: atomic_max_i64((int64_t*)&la[g], iv); break;
      }
      atomicAdd((unsigned long long*)&lc[g], 1ull);
    }
  }
  __syncthreads();
  for (int j = threadIdx.x; j < n_aggs * ngroups; j += blockDim.x) {
    int k = j / ngroups;
    int g = j - k * ngroups;
    int op = aggs[k].op;
    void* acc = aggs[k].acc;
    long long c = lcnt[j];
    if (aggs[k].cnt != nullptr && c)
      atomicAdd((unsigned long long*)&aggs[k].cnt[g], (unsigned long long)c);
    if (op == AGG_SIZE || op == AGG_COUNT) continue;
    if (c == 0) continue;
    switch (op) {
      case AGG_SUM_F64: atomicAdd((double*)acc + g, lacc[j]); break;
      case AGG_SUM_I64: atomicAdd((unsigned long long*)acc + g,
                                  (unsigned long long)__double_as_longlong(lacc[j])); break;
      case AGG_MIN_F64: atomic_min_f64((double*)acc + g, lacc[j]); break;
      case AGG_MAX_F64: atomic_max_f64((double*)acc + g, lacc[j]); break;
      case AGG_MIN_I64: atomic_min_i64((int64_t*)acc + g, __double_as_longlong(lacc[j])); break;
      case AGG_MAX_I64: atomic_max_i64((int64_t*)acc + g, __double_as_longlong(lacc[j])); break;
    }
  }
}

// returns per agg: [acc, cnt?] pairs flattened
std::vector<torch::Tensor> agg_update_fused(
    std::vector<torch::Tensor> datas,
    std::vector<c10::optional<torch::Tensor>> masks,
    std::vector<int64_t> dtypes, std::vector<int64_t> ops,
    std::vector<double> init_fs, std::vector<int64_t> init_is,
    std::vector<int64_t> want_counts,
    torch::Tensor row_gid, int64_t ngroups) {
  auto dev = row_gid.device();
  int64_t n = row_gid.numel();
  int n_aggs = (int)datas.size();
  TORCH_CHECK(n_aggs >= 1 && n_aggs <= MAX_FUSED_AGGS);
  std::vector<torch::Tensor> out;
  FusedAgg fas[MAX_FUSED_AGGS] = {};
  for (int k = 0; k < n_aggs; ++k) {
    int64_t op = ops[k];
    bool f64_acc = (op == AGG_SUM_F64 || op == AGG_MIN_F64 || op == AGG_MAX_F64);
    torch::Tensor acc = f64_acc
        ? torch::full({ngroups}, init_fs[k], torch::dtype(torch::kFloat64).device(dev))
        : torch::full({ngroups}, init_is[k], torch::dtype(torch::kInt64).device(dev));
    torch::Tensor cnt;
    if (want_counts[k] || op == AGG_SIZE || op == AGG_COUNT) {
      cnt = torch::zeros({ngroups}, torch::dtype(torch::kInt64).device(dev));
    }
    fas[k].col = make_desc(datas[k], masks[k], c10::nullopt, c10::nullopt,
                           dtypes[k], n);
    fas[k].op = (int)op;
    fas[k].acc = acc.data_ptr();
    fas[k].cnt = cnt.defined() ? (int64_t*)cnt.data_ptr() : nullptr;
    fas[k].init_f = init_fs[k];
    fas[k].init_i = init_is[k];
    out.push_back(acc);
    out.push_back(cnt.defined() ? cnt : torch::Tensor());
  }
  int block = 256;
  size_t lds_bytes = (size_t)n_aggs * (size_t)ngroups * 16;
  if (lds_bytes <= 48 * 1024) {
    hipLaunchKernelGGL(agg_update_fused_lds_kernel, dim3(grid_for(n, block)),
                       dim3(block), lds_bytes, cur_stream(), fas[0], fas[1],
                       fas[2], fas[3], n_aggs, (int)ngroups,
                       (const int32_t*)row_gid.data_ptr(), n);
  } else {
    hipLaunchKernelGGL(agg_update_fused_kernel, dim3(grid_for(n, block)),
                       dim3(block), 0, cur_stream(), fas[0], fas[1], fas[2],
                       fas[3], n_aggs, (const int32_t*)row_gid.data_ptr(), n);
  }
  CHECK_HIP(hipGetLastError());
  std::vector<torch::Tensor> cleaned;
  for (auto& t : out) cleaned.push_back(t.defined() ? t : torch::empty({0}));
  return cleaned;
}

std::vector<torch::Tensor> agg_update(
    torch::Tensor data, c10::optional<torch::Tensor> mask,
    c10::optional<torch::Tensor> offsets, int64_t dtype,
    torch::Tensor row_gid, int64_t ngroups, int64_t op, double init_f,
    int64_t init_i, bool want_count) {
  auto dev = data.device();
  int64_t n = row_gid.numel();
  torch::Tensor acc;
  bool f64_acc = (op == AGG_SUM_F64 || op == AGG_MIN_F64 || op == AGG_MAX_F64 ||
                  op == AGG_PROD_F64);
  if (f64_acc) {
    acc = torch::full({ngroups}, init_f, torch::dtype(torch::kFloat64).device(dev));
  } else {
    acc = torch::full({ngroups}, init_i, torch::dtype(torch::kInt64).device(dev));
  }
  torch::Tensor cnt;
  int64_t* cnt_ptr = nullptr;
  if (want_count || op == AGG_SIZE || op == AGG_COUNT) {
    cnt = torch::zeros({ngroups}, torch::dtype(torch::kInt64).device(dev));
    cnt_ptr = (int64_t*)cnt.data_ptr();
  }
  ColumnDesc col = make_desc(data, mask, offsets, c10::nullopt, dtype, n);
  int block = 256;
  bool lds_ok = ngroups <= LDS_AGG_MAX_GROUPS && op != AGG_PROD_F64;
  if (lds_ok) {
    hipLaunchKernelGGL(agg_update_lds_kernel, dim3(grid_for(n, block)),
                       dim3(block), 0, cur_stream(), col,
                       (const int32_t*)row_gid.data_ptr(), (int)op,
                       (int)ngroups, acc.data_ptr(), cnt_ptr, n,
                       init_f, init_i);
  } else {
    hipLaunchKernelGGL(agg_update_kernel, dim3(grid_for(n, block)), dim3(block),
                       0, cur_stream(), col,
                       (const int32_t*)row_gid.data_ptr(), (int)op,
                       acc.data_ptr(), cnt_ptr, n);
  }
  CHECK_HIP(hipGetLastError());
  if (cnt.defined()) return {acc, cnt};
  return {acc};
}

// ---------------------------------------------------------------------
// hash join: bucket-chained build table + two-pass probe
// ---------------------------------------------------------------------

__global__ void join_build_kernel(const uint64_t* __restrict__ hashes,
                                  uint32_t* __restrict__ heads,
                                  uint32_t* __restrict__ next,
                                  uint64_t cap_mask, int64_t n) {
  GRID_STRIDE_LOOP(i, n) {
    uint64_t s = hashes[i] & cap_mask;
    uint32_t old = atomicExch(&heads[s], (uint32_t)i);
    next[i] = old;  // old == 0xFFFFFFFF means end of chain
  }
}

// count pass: matches per probe row
__global__ void join_probe_count_kernel(
    const ColumnDesc* __restrict__ bcols, const ColumnDesc* __restrict__ pcols,
    int ncols, const uint64_t* __restrict__ bh, const uint64_t* __restrict__ ph,
    const uint32_t* __restrict__ heads, const uint32_t* __restrict__ next,
    uint64_t cap_mask, int how,  // 0=inner,1=left,2=semi,3=anti
    int64_t* __restrict__ counts, int64_t n_probe) {
  GRID_STRIDE_LOOP(i, n_probe) {
    uint64_t h = ph[i];
    uint32_t cur = heads[h & cap_mask];
    int64_t c = 0;
    while (cur != 0xFFFFFFFFu) {
      if (bh[cur] == h && rows_eq2(pcols, bcols, ncols, i, (int64_t)cur)) {
        ++c;
        if (how == 2 || how == 3) break;  // semi/anti need existence only
      }
      cur = next[cur];
    }
    if (how == 0) counts[i] = c;
    else if (how == 1) counts[i] = c ? c : 1;  // left: null row when no match
    else if (how == 2) counts[i] = c ? 1 : 0;
    else counts[i] = c ? 0 : 1;  // anti
  }
}

__global__ void join_probe_fill_kernel(
    const ColumnDesc* __restrict__ bcols, const ColumnDesc* __restrict__ pcols,
    int ncols, const uint64_t* __restrict__ bh, const uint64_t* __restrict__ ph,
    const uint32_t* __restrict__ heads, const uint32_t* __restrict__ next,
    uint64_t cap_mask, int how, const int64_t* __restrict__ offsets,
    int64_t* __restrict__ out_probe, int64_t* __restrict__ out_build,
    uint8_t* __restrict__ build_matched, int64_t n_probe) {
  GRID_STRIDE_LOOP(i, n_probe) {
    uint64_t h = ph[i];
    uint32_t cur = heads[h & cap_mask];
    int64_t o = offsets[i];
    int64_t c = 0;
    while (cur != 0xFFFFFFFFu) {
      if (bh[cur] == h && rows_eq2(pcols, bcols, ncols, i, (int64_t)cur)) {
        if (how == 0 || how == 1) {
          out_probe[o + c] = i;
          out_build[o + c] = (int64_t)cur;
          if (build_matched) build_matched[cur] = 1;
        } else if (how == 2) {  // semi
          out_probe[o] = i;
        }
        ++c;
        if (how >= 2) break;
      }
      cur = next[cur];
    }
    if (how == 1 && c == 0) {
      out_probe[o] = i;
      out_build[o] = -1;
    }
    if (how == 3 && c == 0) out_probe[o] = i;
  }
}

std::vector<torch::Tensor> join_build(torch::Tensor hashes, int64_t n_build) {
  auto dev = hashes.device();
  int64_t cap = 16;
  while (cap < 2 * std::max<int64_t>(n_build, 1)) cap <<= 1;
  auto heads = torch::full({cap}, (int64_t)-1,
                           torch::dtype(torch::kInt32).device(dev));
  auto next = torch::empty({std::max<int64_t>(n_build, 1)},
                           torch::dtype(torch::kInt32).device(dev));
  if (n_build) {
    int block = 256;
    hipLaunchKernelGGL(join_build_kernel, dim3(grid_for(n_build, block)),
                       dim3(block), 0, cur_stream(),
                       (const uint64_t*)hashes.data_ptr(),
                       (uint32_t*)heads.data_ptr(),
                       (uint32_t*)next.data_ptr(), (uint64_t)(cap - 1),
                       n_build);
    CHECK_HIP(hipGetLastError());
  }
  return {heads, next};
}

std::vector<torch::Tensor> join_probe(
    // build key columns
    std::vector<torch::Tensor> bdatas,
    std::vector<c10::optional<torch::Tensor>> bmasks,
    std::vector<c10::optional<torch::Tensor>> boffsets,
    std::vector<c10::optional<torch::Tensor>> bauxs,
    std::vector<int64_t> bdtypes, int64_t n_build, torch::Tensor bh,
    // probe key columns
    std::vector<torch::Tensor> pdatas,
    std::vector<c10::optional<torch::Tensor>> pmasks,
    std::vector<c10::optional<torch::Tensor>> poffsets,
    std::vector<c10::optional<torch::Tensor>> pauxs,
    std::vector<int64_t> pdtypes, int64_t n_probe, torch::Tensor ph,
    torch::Tensor heads, torch::Tensor next, int64_t how,
    bool track_build_matched) {
  auto dev = bh.device();
  auto bds = build_descset(bdatas, bmasks, boffsets, bauxs, bdtypes, n_build);
  auto pds = build_descset(pdatas, pmasks, poffsets, pauxs, pdtypes, n_probe);
  int64_t cap = heads.numel();
  auto counts = torch::zeros({n_probe}, torch::dtype(torch::kInt64).device(dev));
  int block = 256;
  if (n_probe) {
    hipLaunchKernelGGL(join_probe_count_kernel,
                       dim3(grid_for(n_probe, block)), dim3(block), 0,
                       cur_stream(), bds.ptr(), pds.ptr(), (int)bdatas.size(),
                       (const uint64_t*)bh.data_ptr(),
                       (const uint64_t*)ph.data_ptr(),
                       (const uint32_t*)heads.data_ptr(),
                       (const uint32_t*)next.data_ptr(),
                       (uint64_t)(cap - 1), (int)how,
                       (int64_t*)counts.data_ptr(), n_probe);
    CHECK_HIP(hipGetLastError());
  }
  auto offs = torch::zeros({n_probe + 1}, torch::dtype(torch::kInt64).device(dev));
  auto offs_tail = offs.slice(0, 1, n_probe + 1);
  at::cumsum_out(offs_tail, counts, 0);
  int64_t total = n_probe ? offs[n_probe].item<int64_t>() : 0;
  auto out_probe = torch::empty({total}, torch::dtype(torch::kInt64).device(dev));
  auto out_build = torch::empty({(how == 0 || how == 1) ? total : 0},
                                torch::dtype(torch::kInt64).device(dev));
  torch::Tensor matched;
  uint8_t* matched_ptr = nullptr;
  if (track_build_matched) {
    matched = torch::zeros({std::max<int64_t>(n_build, 1)},
                           torch::dtype(torch::kUInt8).device(dev));
    matched_ptr = (uint8_t*)matched.data_ptr();
  }
  if (n_probe && total) {
    hipLaunchKernelGGL(join_probe_fill_kernel,
                       dim3(grid_for(n_probe, block)), dim3(block), 0,
                       cur_stream(), bds.ptr(), pds.ptr(), (int)bdatas.size(),
                       (const uint64_t*)bh.data_ptr(),
                       (const uint64_t*)ph.data_ptr(),
                       (const uint32_t*)heads.data_ptr(),
                       (const uint32_t*)next.data_ptr(),
                       (uint64_t)(cap - 1), (int)how,
                       (const int64_t*)offs.data_ptr(),
                       (int64_t*)out_probe.data_ptr(),
                       (int64_t*)out_build.data_ptr(), matched_ptr, n_probe);
    CHECK_HIP(hipGetLastError());
  }
  if (track_build_matched) return {out_probe, out_build, matched};
  return {out_probe, out_build};
}

// ---------------------------------------------------------------------
// Parquet RLE/bit-packed hybrid expansion (on-GPU parquet decode path;
// reference role: cudf::io::read_parquet in gpu_read_parquet.h, here a
// hand-written gfx950 kernel).  Run table parsed on host from page bytes:
// kind 0 = RLE run (value repeated count times), kind 1 = bit-packed group
// (count values packed at `bitwidth` starting at bit_offset).

struct RleRun {
  int64_t out_start;
  int32_t count;
  int32_t kind;
  int64_t value_or_bitoff;
};

__global__ void rle_expand_kernel(const RleRun* __restrict__ runs, int n_runs,
                                  const uint8_t* __restrict__ packed,
                                  int bitwidth, int32_t* __restrict__ out) {
  // one wave per run slot, grid-stride over runs; lanes fill values
  int64_t wave_id = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int lane = threadIdx.x % WAVE;
  int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  for (int64_t r = wave_id; r < n_runs; r += n_waves) {
    RleRun run = runs[r];
    if (run.kind == 0) {
      int32_t v = (int32_t)run.value_or_bitoff;
      for (int64_t i = lane; i < run.count; i += WAVE) {
        out[run.out_start + i] = v;
      }
    } else {
      int64_t bit0 = run.value_or_bitoff;
      for (int64_t i = lane; i < run.count; i += WAVE) {
        int64_t bp = bit0 + i * bitwidth;
        int64_t byte = bp >> 3;
        int shift = (int)(bp & 7);
        // up to 32-bit reads cover bitwidth <= 24; parquet dict ids fit
        uint32_t w = (uint32_t)packed[byte]
                     | ((uint32_t)packed[byte + 1] << 8)
                     | ((uint32_t)packed[byte + 2] << 16)
                     | ((uint32_t)packed[byte + 3] << 24);
        out[run.out_start + i] = (int32_t)((w >> shift)
                                           & ((1u << bitwidth) - 1));
      }
    }
  }
}

torch::Tensor rle_expand(torch::Tensor runs_blob, int64_t n_runs,
                         torch::Tensor packed, int64_t bitwidth,
                         int64_t n_out) {
  auto dev = packed.device();
  auto out = torch::empty({n_out}, torch::dtype(torch::kInt32).device(dev));
  if (n_runs) {
    int block = 256;
    int waves_per_block = block / WAVE;
    int grid = (int)std::min<int64_t>(
        (n_runs + waves_per_block - 1) / waves_per_block, 2048);
    hipLaunchKernelGGL(rle_expand_kernel, dim3(grid), dim3(block), 0,
                       cur_stream(), (const RleRun*)runs_blob.data_ptr(),
                       (int)n_runs, (const uint8_t*)packed.data_ptr(),
                       (int)bitwidth, (int32_t*)out.data_ptr());
    CHECK_HIP(hipGetLastError());
  }
  return out;
}

torch::Tensor gemm_f32(torch::Tensor A, torch::Tensor B);  // gemm.hip

// parquet.hip: page-parallel on-GPU parquet decode
void pq_decompress(torch::Tensor src, torch::Tensor pages_blob,
                   int64_t n_pages, torch::Tensor scratch);
std::vector<torch::Tensor> pq_def_levels(torch::Tensor scratch,
                                         torch::Tensor pages_blob,
                                         int64_t n_pages, int64_t bitwidth,
                                         int64_t max_def, int64_t total_nv);
torch::Tensor pq_expand_codes(torch::Tensor scratch, torch::Tensor pages_blob,
                              int64_t n_pages,
                              c10::optional<torch::Tensor> val_data_off,
                              torch::Tensor dense_off,
                              c10::optional<torch::Tensor> n_valid,
                              int64_t dense_total);
torch::Tensor pq_copy_fixed(torch::Tensor scratch, torch::Tensor pages_blob,
                            int64_t n_pages,
                            c10::optional<torch::Tensor> val_data_off,
                            torch::Tensor dense_off,
                            c10::optional<torch::Tensor> n_valid,
                            int64_t esize, int64_t dense_total);
std::vector<torch::Tensor> pq_byte_array_lengths(
    torch::Tensor scratch, torch::Tensor pages_blob, int64_t n_pages,
    c10::optional<torch::Tensor> val_data_off, torch::Tensor dense_off,
    c10::optional<torch::Tensor> n_valid, int64_t dense_total);
torch::Tensor pq_copy_strings(torch::Tensor scratch, torch::Tensor src_abs,
                              torch::Tensor dst_off, torch::Tensor lengths,
                              int64_t n, int64_t total_bytes);
torch::Tensor pq_parse_headers(torch::Tensor chunk_buf);
std::vector<torch::Tensor> gather_multi(std::vector<torch::Tensor> srcs,
                                        torch::Tensor idx);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm_f32", &gemm_f32, "f32 MFMA GEMM (v_mfma_f32_16x16x4_f32)");
  m.def("rle_expand", &rle_expand, "parquet RLE/bit-packed hybrid expand");
  m.def("agg_update_fused", &agg_update_fused, "fused multi-aggregate update");
  m.def("groupby_build_packed", &groupby_build_packed,
        "packed-u64-key hash groupby build");
  m.def("hash_columns", &hash_columns, "multi-column row hash");
  m.def("dt_field", &dt_field, "datetime field extraction");
  m.def("gather_string", &gather_string, "string column gather");
  m.def("groupby_build", &groupby_build, "hash groupby build");
  m.def("agg_update", &agg_update, "aggregate update");
  m.def("join_build", &join_build, "hash join build");
  m.def("join_probe", &join_probe, "hash join probe");
  m.def("pq_decompress", &pq_decompress,
        "page-parallel snappy decompress (one wave/page)");
  m.def("pq_def_levels", &pq_def_levels,
        "definition levels -> validity mask + per-page valid counts");
  m.def("pq_expand_codes", &pq_expand_codes,
        "RLE/bit-packed dictionary codes -> dense int32");
  m.def("pq_copy_fixed", &pq_copy_fixed,
        "PLAIN fixed-width page values -> dense buffer");
  m.def("pq_byte_array_lengths", &pq_byte_array_lengths,
        "PLAIN BYTE_ARRAY length/offset walk");
  m.def("pq_copy_strings", &pq_copy_strings,
        "gather string bytes to packed buffer");
  m.def("pq_parse_headers", &pq_parse_headers,
        "host-side thrift page-header parse of a chunk buffer");
  m.def("gather_multi", &gather_multi,
        "fused multi-column fixed-width gather (one launch per table)");
  m.attr("_native") = true;
}
