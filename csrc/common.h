// Common device definitions for bodo_amd gfx950 kernels.
//
// Written natively for CDNA4 (wave64, 64-wide wavefronts, LDS 160 KiB/CU);
// no CUDA compatibility paths.  Reference behavioral spec:
// bodo/libs/_array_hash.cpp (row hashing), _bodo_common.h (column ABI).
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// Must match bodo_amd/core/types.py TypeKind
enum BodoType : int {
  BT_INT8 = 0, BT_INT16 = 1, BT_INT32 = 2, BT_INT64 = 3,
  BT_FLOAT32 = 4, BT_FLOAT64 = 5, BT_BOOL = 6, BT_DATE32 = 7,
  BT_TIMESTAMP_NS = 8, BT_STRING = 9, BT_DICT = 10, BT_DECIMAL128 = 11,
  BT_UINT8 = 12, BT_UINT16 = 13, BT_UINT32 = 14, BT_UINT64 = 15,
};

// Column view passed to kernels (array-of-structs in device memory).
struct ColumnDesc {
  const void* data;        // fixed-width values / string bytes / dict codes
  const int64_t* offsets;  // STRING: n+1 offsets
  const uint8_t* mask;     // validity (1=valid) or nullptr
  const uint64_t* aux;     // DICT: per-code value-hash LUT
  int dtype;
  int64_t n;
};

// ---------------------------------------------------------------- hashing
// splitmix64-style finalizer; must match ops/__init__.py _mix64_np
DEV_INLINE uint64_t mix64(uint64_t x) {
  x ^= x >> 33;
  x *= 0xff51afd7ed558ccdULL;
  x ^= x >> 33;
  x *= 0xc4ceb9fe1a85ec53ULL;
  x ^= x >> 33;
  return x;
}

#define GOLDEN 0x9E3779B97F4A7C15ULL
#define NULL_HASH GOLDEN

DEV_INLINE uint64_t fnv1a(const uint8_t* p, int64_t len) {
  uint64_t h = 0xcbf29ce484222325ULL;
  for (int64_t i = 0; i < len; ++i) {
    h = (h ^ (uint64_t)p[i]) * 0x100000001b3ULL;
  }
  return h;
}

DEV_INLINE uint64_t f64_bits_norm(double v) {
  if (v == 0.0) return 0;  // -0.0 == 0.0
  if (v != v) return 0x7FF8000000000000ULL;  // canonical NaN
  return __double_as_longlong(v);
}

// hash one row of one column (no seed/mask handling)
DEV_INLINE uint64_t hash_value(const ColumnDesc& c, int64_t i) {
  switch (c.dtype) {
    case BT_INT8: return mix64((uint64_t)(int64_t)((const int8_t*)c.data)[i]);
    case BT_UINT8: return mix64((uint64_t)(int64_t)((const uint8_t*)c.data)[i]);
    case BT_INT16: return mix64((uint64_t)(int64_t)((const int16_t*)c.data)[i]);
    case BT_UINT16: return mix64((uint64_t)(int64_t)((const int16_t*)c.data)[i]);
    case BT_INT32: case BT_DATE32:
      return mix64((uint64_t)(int64_t)((const int32_t*)c.data)[i]);
    case BT_UINT32: return mix64((uint64_t)(int64_t)((const int32_t*)c.data)[i]);
    case BT_INT64: case BT_TIMESTAMP_NS: case BT_UINT64: case BT_DECIMAL128:
      // DECIMAL128 is stored as scaled int64 (exact for p <= 18)
      return mix64((uint64_t)((const int64_t*)c.data)[i]);
    case BT_BOOL: return mix64((uint64_t)((const uint8_t*)c.data)[i]);
    case BT_FLOAT32: return mix64(f64_bits_norm((double)((const float*)c.data)[i]));
    case BT_FLOAT64: return mix64(f64_bits_norm(((const double*)c.data)[i]));
    case BT_STRING: {
      int64_t s = c.offsets[i], e = c.offsets[i + 1];
      return fnv1a((const uint8_t*)c.data + s, e - s);
    }
    case BT_DICT: {
      int32_t code = ((const int32_t*)c.data)[i];
      return c.aux[code];
    }
  }
  return 0;
}

DEV_INLINE bool is_valid_at(const ColumnDesc& c, int64_t i) {
  return c.mask == nullptr || c.mask[i];
}

// null-aware equality of row a and row b of the same column set
DEV_INLINE bool value_eq(const ColumnDesc& c, int64_t a, int64_t b) {
  bool va = is_valid_at(c, a), vb = is_valid_at(c, b);
  if (va != vb) return false;
  if (!va) return true;  // both null
  switch (c.dtype) {
    case BT_INT8: case BT_UINT8: case BT_BOOL:
      return ((const int8_t*)c.data)[a] == ((const int8_t*)c.data)[b];
    case BT_INT16: case BT_UINT16:
      return ((const int16_t*)c.data)[a] == ((const int16_t*)c.data)[b];
    case BT_INT32: case BT_UINT32: case BT_DATE32: case BT_DICT:
      return ((const int32_t*)c.data)[a] == ((const int32_t*)c.data)[b];
    case BT_INT64: case BT_UINT64: case BT_TIMESTAMP_NS: case BT_DECIMAL128:
      return ((const int64_t*)c.data)[a] == ((const int64_t*)c.data)[b];
    case BT_FLOAT32: {
      float x = ((const float*)c.data)[a], y = ((const float*)c.data)[b];
      return (x == y) || (x != x && y != y);
    }
    case BT_FLOAT64: {
      double x = ((const double*)c.data)[a], y = ((const double*)c.data)[b];
      return (x == y) || (x != x && y != y);
    }
    case BT_STRING: {
      int64_t sa = c.offsets[a], ea = c.offsets[a + 1];
      int64_t sb = c.offsets[b], eb = c.offsets[b + 1];
      if (ea - sa != eb - sb) return false;
      const uint8_t* pa = (const uint8_t*)c.data + sa;
      const uint8_t* pb = (const uint8_t*)c.data + sb;
      for (int64_t k = 0; k < ea - sa; ++k)
        if (pa[k] != pb[k]) return false;
      return true;
    }
  }
  return false;
}

// equality of row a in columns ca[] vs row b in columns cb[] (join probe)
DEV_INLINE bool rows_eq2(const ColumnDesc* ca, const ColumnDesc* cb, int ncols,
                         int64_t a, int64_t b) {
  for (int k = 0; k < ncols; ++k) {
    const ColumnDesc& x = ca[k];
    const ColumnDesc& y = cb[k];
    bool va = is_valid_at(x, a), vb = is_valid_at(y, b);
    if (va != vb) return false;
    if (!va) continue;  // both null on this key: equal, check next
    switch (x.dtype) {
      case BT_INT8: case BT_UINT8: case BT_BOOL:
        if (((const int8_t*)x.data)[a] != ((const int8_t*)y.data)[b]) return false;
        break;
      case BT_INT16: case BT_UINT16:
        if (((const int16_t*)x.data)[a] != ((const int16_t*)y.data)[b]) return false;
        break;
      case BT_INT32: case BT_UINT32: case BT_DATE32: case BT_DICT:
        if (((const int32_t*)x.data)[a] != ((const int32_t*)y.data)[b]) return false;
        break;
      case BT_INT64: case BT_UINT64: case BT_TIMESTAMP_NS: case BT_DECIMAL128:
        if (((const int64_t*)x.data)[a] != ((const int64_t*)y.data)[b]) return false;
        break;
      case BT_FLOAT32: {
        float u = ((const float*)x.data)[a], v = ((const float*)y.data)[b];
        if (!((u == v) || (u != u && v != v))) return false;
        break;
      }
      case BT_FLOAT64: {
        double u = ((const double*)x.data)[a], v = ((const double*)y.data)[b];
        if (!((u == v) || (u != u && v != v))) return false;
        break;
      }
      case BT_STRING: {
        int64_t sa = x.offsets[a], ea = x.offsets[a + 1];
        int64_t sb = y.offsets[b], eb = y.offsets[b + 1];
        if (ea - sa != eb - sb) return false;
        const uint8_t* pa = (const uint8_t*)x.data + sa;
        const uint8_t* pb = (const uint8_t*)y.data + sb;
        for (int64_t t = 0; t < ea - sa; ++t)
          if (pa[t] != pb[t]) return false;
        break;
      }
      default:
        return false;
    }
  }
  return true;
}

DEV_INLINE bool rows_eq(const ColumnDesc* cols, int ncols, int64_t a, int64_t b) {
  for (int k = 0; k < ncols; ++k)
    if (!value_eq(cols[k], a, b)) return false;
  return true;
}

// ---------------------------------------------------------------- atomics
DEV_INLINE void atomic_min_f64(double* addr, double val) {
  unsigned long long* a = (unsigned long long*)addr;
  unsigned long long old = *a;
  while (true) {
    double cur = __longlong_as_double(old);
    if (!(val < cur)) break;
    unsigned long long prev =
        atomicCAS(a, old, (unsigned long long)__double_as_longlong(val));
    if (prev == old) break;
    old = prev;
  }
}

DEV_INLINE void atomic_max_f64(double* addr, double val) {
  unsigned long long* a = (unsigned long long*)addr;
  unsigned long long old = *a;
  while (true) {
    double cur = __longlong_as_double(old);
    if (!(val > cur)) break;
    unsigned long long prev =
        atomicCAS(a, old, (unsigned long long)__double_as_longlong(val));
    if (prev == old) break;
    old = prev;
  }
}

DEV_INLINE void atomic_min_i64(int64_t* addr, int64_t val) {
  unsigned long long* a = (unsigned long long*)addr;
  unsigned long long old = *a;
  while (true) {
    if (!(val < (int64_t)old)) break;
    unsigned long long prev = atomicCAS(a, old, (unsigned long long)val);
    if (prev == old) break;
    old = prev;
  }
}

DEV_INLINE void atomic_max_i64(int64_t* addr, int64_t val) {
  unsigned long long* a = (unsigned long long*)addr;
  unsigned long long old = *a;
  while (true) {
    if (!(val > (int64_t)old)) break;
    unsigned long long prev = atomicCAS(a, old, (unsigned long long)val);
    if (prev == old) break;
    old = prev;
  }
}

// grid sizing: memory-bound ops cap at ~2048 blocks and grid-stride
// (guide §6 Guideline 11)
inline int grid_for(int64_t n, int block, int max_blocks = 2048) {
  int64_t b = (n + block - 1) / block;
  if (b > max_blocks) b = max_blocks;
  if (b < 1) b = 1;
  return (int)b;
}

#define GRID_STRIDE_LOOP(i, n)                                    \
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; \
       i < (n); i += (int64_t)gridDim.x * blockDim.x)
