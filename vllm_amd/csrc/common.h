// Common device helpers for vllm_amd CDNA4 (gfx950) kernels.
//
// Written for MI355X only: wave64, 32-bank LDS, MFMA matrix cores,
// HBM3E. No CUDA-compat paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#include <cstdint>

#define WAVE_SIZE 64

#define DEVINL __device__ __forceinline__

// ---------------------------------------------------------------------------
// Vector types for wide loads (G13: always vectorize bf16 loads).
typedef short  s16x8 __attribute__((ext_vector_type(8)));   // 8 bf16 = 16B
typedef short  s16x4 __attribute__((ext_vector_type(4)));   // 8B
typedef float  f32x4 __attribute__((ext_vector_type(4)));
typedef float  f32x2 __attribute__((ext_vector_type(2)));
typedef float  f32x16 __attribute__((ext_vector_type(16)));
typedef int    i32x4 __attribute__((ext_vector_type(4)));

// MFMA fragment types (gfx950 16x16x32 bf16: A/B = 8 bf16, C/D = 4 f32).
typedef s16x8 frag_b16;
typedef f32x4 frag_f32;

DEVINL float bf16_to_f32(short v) {
  union {
    float f;
    uint32_t u;
  } cvt;
  cvt.u = (uint32_t)(uint16_t)v << 16;
  return cvt.f;
}

DEVINL short f32_to_bf16(float f) {
  union {
    float f;
    uint32_t u;
  } cvt;
  cvt.f = f;
  // round-to-nearest-even
  uint32_t lsb = (cvt.u >> 16) & 1;
  uint32_t rounded = cvt.u + 0x7fffu + lsb;
  return (short)(rounded >> 16);
}

DEVINL float f16_to_f32(short v) { return __half2float(__ushort_as_half((uint16_t)v)); }
DEVINL short f16_from_f32(float f) { return (short)__half_as_ushort(__float2half(f)); }

// Generic scalar converters selected by template tag.
struct BF16Tag {};
struct FP16Tag {};

template <typename Tag> DEVINL float to_f32(short v);
template <> DEVINL float to_f32<BF16Tag>(short v) { return bf16_to_f32(v); }
template <> DEVINL float to_f32<FP16Tag>(short v) { return f16_to_f32(v); }

template <typename Tag> DEVINL short from_f32(float f);
template <> DEVINL short from_f32<BF16Tag>(float f) { return f32_to_bf16(f); }
template <> DEVINL short from_f32<FP16Tag>(float f) { return f16_from_f32(f); }

// ---------------------------------------------------------------------------
// OCP fp8 e4m3fn (bias 7, no inf, 0x7f/0xff = NaN, max finite 448) — the
// gfx950-native FP8 format and torch.float8_e4m3fn. Used for the fp8 KV
// cache (kv_cache_dtype="fp8"): halves attention HBM traffic and doubles
// KV capacity.
DEVINL float fp8_e4m3_to_f32(uint8_t v) {
  const uint32_t sign = (uint32_t)(v & 0x80) << 24;
  const uint32_t e = (v >> 3) & 0xf;
  const uint32_t m = v & 0x7;
  uint32_t u;
  if (e == 0) {
    // subnormal: m * 2^-9
    if (m == 0) {
      u = sign;
    } else {
      // normalize
      // m = 1.xx * 2^(2-shift); value = m * 2^-9 = 1.xx * 2^(-7-shift)
      int shift = (m & 4) ? 0 : ((m & 2) ? 1 : 2);
      uint32_t mant = (m << (shift + 1)) & 0x7;  // drop leading 1
      u = sign | ((uint32_t)(120 - shift) << 23) | (mant << 20);
    }
  } else if (e == 0xf && m == 0x7) {
    u = sign | 0x7fc00000u;  // NaN
  } else {
    u = sign | ((e - 7 + 127) << 23) | (m << 20);
  }
  union { uint32_t u; float f; } cvt;
  cvt.u = u;
  return cvt.f;
}

DEVINL uint8_t f32_to_fp8_e4m3(float f) {
  union { float f; uint32_t u; } cvt;
  cvt.f = f;
  const uint8_t sign = (uint8_t)((cvt.u >> 24) & 0x80);
  const float af = fabsf(f);
  if (af != af) return sign | 0x7f;
  if (af >= 464.f) return sign | 0x7e;  // saturate to 448 (RNE boundary)
  cvt.f = af;
  const int32_t e32 = (int32_t)((cvt.u >> 23) & 0xff);
  const uint32_t m32 = cvt.u & 0x7fffffu;
  int32_t e8 = e32 - 127 + 7;
  if (e8 >= 1) {
    // normal range
    uint32_t mant = m32 >> 20;
    const uint32_t rem = m32 & 0xfffffu;
    if (rem > 0x80000u || (rem == 0x80000u && (mant & 1))) mant++;
    if (mant == 8) { mant = 0; e8++; }
    return sign | (uint8_t)((e8 << 3) | mant);
  }
  // subnormal: value = mant_full * 2^(e32-127-23); quantum 2^-9.
  if (af < 0.0009765625f) return sign;  // < 2^-10 -> rounds to 0
  const uint32_t full = 0x800000u | m32;          // 24-bit mantissa
  const int shift = 20 + (1 - e8);                // bits to drop
  uint32_t mant = full >> shift;
  const uint32_t rem = full & ((1u << shift) - 1);
  const uint32_t half = 1u << (shift - 1);
  if (rem > half || (rem == half && (mant & 1))) mant++;
  if (mant >= 8) return sign | 0x08;  // rounded up into normal min
  return sign | (uint8_t)mant;
}

// Cache element traits: the KV cache may be narrower than the compute
// dtype. vec8 loads 8 cache elements (16 B bf16/fp16, 8 B fp8).
typedef uint8_t u8x8 __attribute__((ext_vector_type(8)));
struct FP8CacheTag {};

template <typename CTag> struct CacheTraits;
template <> struct CacheTraits<BF16Tag> {
  using elem = short;
  using vec8 = s16x8;
  static DEVINL float get(vec8 v, int j) { return bf16_to_f32(v[j]); }
  static DEVINL elem put(float f) { return f32_to_bf16(f); }
};
template <> struct CacheTraits<FP16Tag> {
  using elem = short;
  using vec8 = s16x8;
  static DEVINL float get(vec8 v, int j) { return f16_to_f32(v[j]); }
  static DEVINL elem put(float f) { return f16_from_f32(f); }
};
template <> struct CacheTraits<FP8CacheTag> {
  using elem = uint8_t;
  using vec8 = u8x8;
  static DEVINL float get(vec8 v, int j) { return fp8_e4m3_to_f32(v[j]); }
  static DEVINL elem put(float f) { return f32_to_fp8_e4m3(f); }
};

// ---------------------------------------------------------------------------
// Wave reductions (64 lanes).
DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEVINL float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Reduce within 16-lane groups (used for MFMA row reductions).
DEVINL float group16_reduce_max(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

DEVINL float group16_reduce_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// Block reduction via LDS (block size must be multiple of 64).
template <int MAX_WAVES>
DEVINL float block_reduce_sum(float v, float* lds_scratch) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wave] = v;
  __syncthreads();
  float r = (lane < nwaves) ? lds_scratch[lane] : 0.f;
#pragma unroll
  for (int off = MAX_WAVES / 2; off > 0; off >>= 1)
    r += __shfl_xor(r, off, 64);
  return __shfl(r, 0, 64);
}

// ---------------------------------------------------------------------------
#define HIP_CHECK_KERNEL()                                    \
  do {                                                        \
    hipError_t e = hipGetLastError();                         \
    if (e != hipSuccess) {                                    \
      TORCH_CHECK(false, "HIP kernel launch failed: ",        \
                  hipGetErrorString(e));                      \
    }                                                         \
  } while (0)

DEVINL int ceil_div_dev(int a, int b) { return (a + b - 1) / b; }
inline int ceil_div(int a, int b) { return (a + b - 1) / b; }
