// torchmpi_amd core definitions shared by the CPU transport and GPU engine.
//
// Re-designs (MI355X-native, from scratch) the role of TorchMPI's
// lib/collectives.h type plumbing without any MPI.
#pragma once

#include <cstddef>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace tma {

// Reduction data types supported across both transports.
enum class DType : int32_t {
  F32 = 0,
  F64 = 1,
  F16 = 2,
  BF16 = 3,
  I32 = 4,
  I64 = 5,
  U8 = 6,
  I8 = 7,
};

enum class RedOp : int32_t {
  SUM = 0,
  MAX = 1,
  MIN = 2,
  PROD = 3,
};

inline size_t dtype_size(DType d) {
  switch (d) {
    case DType::F32: return 4;
    case DType::F64: return 8;
    case DType::F16: return 2;
    case DType::BF16: return 2;
    case DType::I32: return 4;
    case DType::I64: return 8;
    case DType::U8: return 1;
    case DType::I8: return 1;
  }
  throw std::runtime_error("torchmpi_amd: unknown dtype");
}

#define TMA_CHECK(cond, msg)                                       \
  do {                                                             \
    if (!(cond)) {                                                 \
      throw std::runtime_error(std::string("torchmpi_amd: ") + msg); \
    }                                                              \
  } while (0)

// ---------------------------------------------------------------------------
// CPU-side elementwise reduction (the TCP transport's local combine step).
// ---------------------------------------------------------------------------

inline float bf16_to_f32(uint16_t v) {
  uint32_t u = static_cast<uint32_t>(v) << 16;
  float f;
  std::memcpy(&f, &u, 4);
  return f;
}

inline uint16_t f32_to_bf16(float f) {
  uint32_t u;
  std::memcpy(&u, &f, 4);
  // round-to-nearest-even
  uint32_t rounding = 0x7fff + ((u >> 16) & 1);
  u += rounding;
  return static_cast<uint16_t>(u >> 16);
}

inline float f16_to_f32(uint16_t h) {
  uint32_t sign = (h >> 15) & 1, exp = (h >> 10) & 0x1f, man = h & 0x3ff;
  uint32_t u;
  if (exp == 0) {
    if (man == 0) {
      u = sign << 31;
    } else {
      // subnormal
      int e = -1;
      uint32_t m = man;
      while (!(m & 0x400)) { m <<= 1; --e; }
      m &= 0x3ff;
      u = (sign << 31) | ((127 - 15 + e + 1) << 23) | (m << 13);
    }
  } else if (exp == 31) {
    u = (sign << 31) | 0x7f800000 | (man << 13);
  } else {
    u = (sign << 31) | ((exp - 15 + 127) << 23) | (man << 13);
  }
  float f;
  std::memcpy(&f, &u, 4);
  return f;
}

inline uint16_t f32_to_f16(float f) {
  uint32_t u;
  std::memcpy(&u, &f, 4);
  uint32_t sign = (u >> 16) & 0x8000;
  int32_t exp = static_cast<int32_t>((u >> 23) & 0xff) - 127 + 15;
  uint32_t man = u & 0x7fffff;
  if (((u >> 23) & 0xff) == 0xff) return static_cast<uint16_t>(sign | 0x7c00 | (man ? 0x200 : 0));
  if (exp >= 31) return static_cast<uint16_t>(sign | 0x7c00);  // overflow -> inf
  if (exp <= 0) {
    if (exp < -10) return static_cast<uint16_t>(sign);
    man |= 0x800000;
    uint32_t shift = static_cast<uint32_t>(14 - exp);
    uint32_t half = man >> shift;
    uint32_t rem = man & ((1u << shift) - 1);
    half += (rem > (1u << (shift - 1))) || ((rem == (1u << (shift - 1))) && (half & 1));
    return static_cast<uint16_t>(sign | half);
  }
  uint32_t half = (static_cast<uint32_t>(exp) << 10) | (man >> 13);
  uint32_t rem = man & 0x1fff;
  half += (rem > 0x1000) || ((rem == 0x1000) && (half & 1));
  return static_cast<uint16_t>(sign | half);
}

template <typename T>
inline void reduce_loop(T* dst, const T* src, size_t n, RedOp op) {
  switch (op) {
    case RedOp::SUM:
      for (size_t i = 0; i < n; ++i) dst[i] = dst[i] + src[i];
      break;
    case RedOp::MAX:
      for (size_t i = 0; i < n; ++i) dst[i] = dst[i] > src[i] ? dst[i] : src[i];
      break;
    case RedOp::MIN:
      for (size_t i = 0; i < n; ++i) dst[i] = dst[i] < src[i] ? dst[i] : src[i];
      break;
    case RedOp::PROD:
      for (size_t i = 0; i < n; ++i) dst[i] = dst[i] * src[i];
      break;
  }
}

// 16-bit float types reduce through f32 accumulation.
template <float (*ToF)(uint16_t), uint16_t (*FromF)(float)>
inline void reduce_loop16(uint16_t* dst, const uint16_t* src, size_t n, RedOp op) {
  for (size_t i = 0; i < n; ++i) {
    float a = ToF(dst[i]), b = ToF(src[i]), r;
    switch (op) {
      case RedOp::SUM: r = a + b; break;
      case RedOp::MAX: r = a > b ? a : b; break;
      case RedOp::MIN: r = a < b ? a : b; break;
      default: r = a * b; break;
    }
    dst[i] = FromF(r);
  }
}

inline void cpu_reduce(void* dst, const void* src, size_t count, DType d, RedOp op) {
  switch (d) {
    case DType::F32:
      reduce_loop(static_cast<float*>(dst), static_cast<const float*>(src), count, op);
      break;
    case DType::F64:
      reduce_loop(static_cast<double*>(dst), static_cast<const double*>(src), count, op);
      break;
    case DType::I32:
      reduce_loop(static_cast<int32_t*>(dst), static_cast<const int32_t*>(src), count, op);
      break;
    case DType::I64:
      reduce_loop(static_cast<int64_t*>(dst), static_cast<const int64_t*>(src), count, op);
      break;
    case DType::U8:
      reduce_loop(static_cast<uint8_t*>(dst), static_cast<const uint8_t*>(src), count, op);
      break;
    case DType::I8:
      reduce_loop(static_cast<int8_t*>(dst), static_cast<const int8_t*>(src), count, op);
      break;
    case DType::BF16:
      reduce_loop16<bf16_to_f32, f32_to_bf16>(
          static_cast<uint16_t*>(dst), static_cast<const uint16_t*>(src), count, op);
      break;
    case DType::F16:
      reduce_loop16<f16_to_f32, f32_to_f16>(
          static_cast<uint16_t*>(dst), static_cast<const uint16_t*>(src), count, op);
      break;
  }
}

// Split `count` elements into `parts` nearly-equal contiguous segments.
// Segment i covers [offsets[i], offsets[i] + counts[i]).
inline void split_segments(size_t count, int parts, std::vector<size_t>& offsets,
                           std::vector<size_t>& counts) {
  offsets.resize(parts);
  counts.resize(parts);
  size_t base = count / parts, rem = count % parts, off = 0;
  for (int i = 0; i < parts; ++i) {
    counts[i] = base + (static_cast<size_t>(i) < rem ? 1 : 0);
    offsets[i] = off;
    off += counts[i];
  }
}

}  // namespace tma
