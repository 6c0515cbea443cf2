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
// gfx950 has single-instruction OCP fp8 converts (v_cvt_f32_fp8 /
// v_cvt_pk_fp8_f32, RNE) — the software bit manipulation this replaced
// cost ~20 VALU ops per element and made the fp8-KV decode kernel
// conversion-bound instead of HBM-bound.
DEVINL float fp8_e4m3_to_f32(uint8_t v) {
  return __builtin_amdgcn_cvt_f32_fp8((uint32_t)v, 0);
}

DEVINL uint8_t f32_to_fp8_e4m3(float f) {
  union { float f; uint32_t u; } c;
  c.f = f;
  // Saturate finite values to +-448 (max finite e4m3); NaN propagates
  // through the hardware convert (same recipe as amd_hip_fp8.h).
  float cl = f;
  if ((c.u & 0x7F800000u) != 0x7F800000u)
    cl = __builtin_amdgcn_fmed3f(f, 448.f, -448.f);
  return (uint8_t)__builtin_amdgcn_cvt_pk_fp8_f32(cl, cl, 0, false);
}

// Packed pair converts for vectorized cache reads: 2 elements/instr.
DEVINL f32x2 fp8x2_e4m3_to_f32x2(uint16_t v2) {
  return __builtin_amdgcn_cvt_pk_f32_fp8((uint32_t)v2, false);
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
