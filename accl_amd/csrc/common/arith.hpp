// Elementwise arithmetic + dtype conversion ("compression lane"), shared
// host/device. This is the semantic analogue of the reference's arith plugin
// (reference: kernels/plugins/reduce_ops/reduce_ops.cpp:31-107 — SUM/MAX over
// f16/f32/f64/i32/i64) and hp_compression (kernels/plugins/hp_compression/
// hp_compression.cpp:72-144 — f32<->f16 cast lanes); bf16 added for MI355X.
//
// Scalar reference implementations only — the GPU engine's movers override the
// hot dtypes with vectorized packed-math paths (gpu/engine.hip); the CPU
// emulator and tail handling use these.
#pragma once
#include "types.hpp"

namespace accl {

// ---- fp16 / bf16 <-> fp32 bit conversions (portable, host+device) ----------
ACCL_HD inline float half_to_float(u16 h) {
  u32 sign = (u32(h) & 0x8000u) << 16;
  u32 exp = (h >> 10) & 0x1F;
  u32 man = h & 0x3FF;
  u32 bits;
  if (exp == 0) {
    if (man == 0) {
      bits = sign;
    } else {  // subnormal: normalize
      int e = -1;
      u32 m = man;
      do { m <<= 1; ++e; } while (!(m & 0x400));
      bits = sign | ((127 - 15 - e) << 23) | ((m & 0x3FF) << 13);
    }
  } else if (exp == 31) {
    bits = sign | 0x7F800000u | (man << 13);
  } else {
    bits = sign | ((exp - 15 + 127) << 23) | (man << 13);
  }
  float f;
  __builtin_memcpy(&f, &bits, 4);
  return f;
}

ACCL_HD inline u16 float_to_half(float f) {
  u32 x;
  __builtin_memcpy(&x, &f, 4);
  u32 sign = (x >> 16) & 0x8000u;
  i32 exp = i32((x >> 23) & 0xFF) - 127 + 15;
  u32 man = x & 0x7FFFFFu;
  if (((x >> 23) & 0xFF) == 0xFF) return u16(sign | 0x7C00 | (man ? 0x200 : 0));
  if (exp >= 31) return u16(sign | 0x7C00);  // overflow -> inf
  if (exp <= 0) {
    if (exp < -10) return u16(sign);
    man |= 0x800000u;
    u32 shift = u32(14 - exp);
    u32 half_man = man >> shift;
    u32 rem = man & ((1u << shift) - 1);
    u32 halfway = 1u << (shift - 1);
    if (rem > halfway || (rem == halfway && (half_man & 1))) half_man++;
    return u16(sign | half_man);
  }
  u32 half_man = man >> 13;
  u32 rem = man & 0x1FFF;
  if (rem > 0x1000 || (rem == 0x1000 && (half_man & 1))) {
    half_man++;
    if (half_man == 0x400) { half_man = 0; exp++; if (exp >= 31) return u16(sign | 0x7C00); }
  }
  return u16(sign | (u32(exp) << 10) | half_man);
}

ACCL_HD inline float bf16_to_float(u16 h) {
  u32 bits = u32(h) << 16;
  float f;
  __builtin_memcpy(&f, &bits, 4);
  return f;
}

ACCL_HD inline u16 float_to_bf16(float f) {
  u32 x;
  __builtin_memcpy(&x, &f, 4);
  if ((x & 0x7F800000u) == 0x7F800000u && (x & 0x7FFFFFu)) return u16((x >> 16) | 0x40);  // quiet NaN
  u32 lsb = (x >> 16) & 1;
  x += 0x7FFFu + lsb;  // round-to-nearest-even
  return u16(x >> 16);
}

// ---- generic scalar load/store through a DataType tag ----------------------
ACCL_HD inline double load_as_f64(const void* p, u64 i, DataType dt) {
  switch (dt) {
    case DataType::float32: return double(((const float*)p)[i]);
    case DataType::float64: return ((const double*)p)[i];
    case DataType::float16: return double(half_to_float(((const u16*)p)[i]));
    case DataType::bfloat16: return double(bf16_to_float(((const u16*)p)[i]));
    case DataType::int32: return double(((const i32*)p)[i]);
    case DataType::int64: return double(((const i64*)p)[i]);
    case DataType::int8: return double(((const signed char*)p)[i]);
    default: return 0.0;
  }
}

ACCL_HD inline void store_from_f64(void* p, u64 i, DataType dt, double v) {
  switch (dt) {
    case DataType::float32: ((float*)p)[i] = float(v); break;
    case DataType::float64: ((double*)p)[i] = v; break;
    case DataType::float16: ((u16*)p)[i] = float_to_half(float(v)); break;
    case DataType::bfloat16: ((u16*)p)[i] = float_to_bf16(float(v)); break;
    case DataType::int32: ((i32*)p)[i] = i32(v); break;
    case DataType::int64: ((i64*)p)[i] = i64(v); break;
    case DataType::int8: ((signed char*)p)[i] = (signed char)(v); break;
    default: break;
  }
}

// Integer-exact paths: i64 must not round-trip through double.
ACCL_HD inline bool dtype_is_int(DataType dt) {
  return dt == DataType::int32 || dt == DataType::int64 || dt == DataType::int8;
}

// dst[i] = convert(src[i]); src/dst may be the same dtype (plain copy).
ACCL_HD inline void convert_range(const void* src, DataType sdt, void* dst,
                                  DataType ddt, u64 n) {
  if (sdt == ddt) {
    const unsigned char* s = (const unsigned char*)src;
    unsigned char* d = (unsigned char*)dst;
    u64 bytes = n * dtype_size(sdt);
    for (u64 i = 0; i < bytes; ++i) d[i] = s[i];
    return;
  }
  if (sdt == DataType::int64 && ddt == DataType::int64) return;  // unreachable
  for (u64 i = 0; i < n; ++i) {
    if (dtype_is_int(sdt) && dtype_is_int(ddt)) {
      i64 v = (sdt == DataType::int64) ? ((const i64*)src)[i]
              : (sdt == DataType::int32) ? i64(((const i32*)src)[i])
                                         : i64(((const signed char*)src)[i]);
      if (ddt == DataType::int64) ((i64*)dst)[i] = v;
      else if (ddt == DataType::int32) ((i32*)dst)[i] = i32(v);
      else ((signed char*)dst)[i] = (signed char)v;
    } else {
      store_from_f64(dst, i, ddt, load_as_f64(src, i, sdt));
    }
  }
}

// dst[i] = f(a[i], b[i]) — the reduce_ops analogue. All three dtypes may
// differ (compressed operands); math is done in the widest of the inputs.
ACCL_HD inline void combine_range(ReduceFunction f, const void* a, DataType adt,
                                  const void* b, DataType bdt, void* dst,
                                  DataType ddt, u64 n) {
  const bool ints = dtype_is_int(adt) && dtype_is_int(bdt) && dtype_is_int(ddt);
  for (u64 i = 0; i < n; ++i) {
    if (ints) {
      i64 x = (adt == DataType::int64) ? ((const i64*)a)[i]
              : (adt == DataType::int32) ? i64(((const i32*)a)[i])
                                         : i64(((const signed char*)a)[i]);
      i64 y = (bdt == DataType::int64) ? ((const i64*)b)[i]
              : (bdt == DataType::int32) ? i64(((const i32*)b)[i])
                                         : i64(((const signed char*)b)[i]);
      i64 r = (f == ReduceFunction::SUM) ? (x + y) : (x > y ? x : y);
      if (ddt == DataType::int64) ((i64*)dst)[i] = r;
      else if (ddt == DataType::int32) ((i32*)dst)[i] = i32(r);
      else ((signed char*)dst)[i] = (signed char)r;
    } else {
      double x = load_as_f64(a, i, adt);
      double y = load_as_f64(b, i, bdt);
      double r = (f == ReduceFunction::SUM) ? (x + y) : (x > y ? x : y);
      store_from_f64(dst, i, ddt, r);
    }
  }
}

}  // namespace accl
