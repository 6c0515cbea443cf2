// fp8 (OCP e4m3) dynamic activation quantization + GEMM output rescale.
//
// Role of the reference's csrc/quantization/fp8/common.cu
// (dynamic_per_token_scaled_fp8_quant) and the w8a8 scaled-mm epilogue:
// per-token activation scales computed on the fly, weights quantized
// per-output-channel at load, and the fp8 x fp8 -> bf16 hipBLASLt GEMM
// rescaled afterwards by rs[m] * cs[n] in one fused pass (which also
// adds the bias, so the fp8 path needs no hipBLASLt epilogue).
//
// CDNA4 notes: one workgroup per token row for the quant kernel (rows
// are independent; M rows x 256 lanes fills the chip for any decode
// batch), vec8 16-byte bf16 loads, wave64 + LDS block reduction for the
// absmax. E4M3 max finite is 448; scales clamp at 1e-12 so zero rows
// stay representable.

#include <torch/all.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace vllm_amd {

namespace {

constexpr float kFp8Max = 448.f;

template <typename Tag, int BLOCK>
__global__ void dynamic_quant_fp8_kernel(
    uint8_t* __restrict__ out,       // [M, K] e4m3
    float* __restrict__ scales,      // [M]
    const short* __restrict__ x,     // [M, K] bf16/fp16
    const int K) {
  const int row = blockIdx.x;
  const short* xr = x + (int64_t)row * K;
  uint8_t* yr = out + (int64_t)row * K;

  // Pass 1: row absmax (vec8 loads).
  float amax = 0.f;
  for (int i = threadIdx.x * 8; i < K; i += BLOCK * 8) {
    s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) amax = fmaxf(amax, fabsf(to_f32<Tag>(v[j])));
  }
  __shared__ float lds[BLOCK / WAVE_SIZE];
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  amax = wave_reduce_max(amax);
  if (lane == 0) lds[wave] = amax;
  __syncthreads();
  amax = (lane < BLOCK / WAVE_SIZE) ? lds[lane] : 0.f;
  amax = wave_reduce_max(amax);
  amax = __shfl(amax, 0, 64);

  const float scale = fmaxf(amax, 1e-12f) / kFp8Max;
  const float inv = 1.f / scale;
  if (threadIdx.x == 0) scales[row] = scale;

  // Pass 2: convert (row is hot in L2 from pass 1).
  for (int i = threadIdx.x * 8; i < K; i += BLOCK * 8) {
    s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
    u8x8 q;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      q[j] = f32_to_fp8_e4m3(to_f32<Tag>(v[j]) * inv);
    *reinterpret_cast<u8x8*>(yr + i) = q;
  }
}

// y[m, n] = y[m, n] * rs[m] * cs[n] (+ bias[n]) — fused de-scale of the
// raw fp8 GEMM output, bf16 in place.
template <int BLOCK>
__global__ void scale_rows_cols_kernel(
    short* __restrict__ y,             // [M, N] bf16
    const float* __restrict__ rs,      // [M]
    const float* __restrict__ cs,      // [N]
    const short* __restrict__ bias,    // [N] or nullptr
    const int N) {
  const int row = blockIdx.y;
  const float r = rs[row];
  short* yr = y + (int64_t)row * N;
  const int i = (blockIdx.x * BLOCK + threadIdx.x) * 8;
  if (i >= N) return;
  s16x8 v = *reinterpret_cast<const s16x8*>(yr + i);
  s16x8 b;
  if (bias != nullptr) b = *reinterpret_cast<const s16x8*>(bias + i);
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float f = bf16_to_f32(v[j]) * r * cs[i + j];
    if (bias != nullptr) f += bf16_to_f32(b[j]);
    v[j] = f32_to_bf16(f);
  }
  *reinterpret_cast<s16x8*>(yr + i) = v;
}

}  // namespace

// x: [M, K] bf16/fp16 -> (out [M, K] float8_e4m3fn, scales [M] f32).
void dynamic_quant_fp8(torch::Tensor out, torch::Tensor scales,
                       torch::Tensor x) {
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  const int M = x.size(0);
  const int K = x.size(1);
  TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8, got ", K);
  constexpr int BLOCK = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  if (bf16) {
    hipLaunchKernelGGL((dynamic_quant_fp8_kernel<BF16Tag, BLOCK>), dim3(M),
                       dim3(BLOCK), 0, stream,
                       (uint8_t*)out.data_ptr(), scales.data_ptr<float>(),
                       (const short*)x.data_ptr(), K);
  } else {
    hipLaunchKernelGGL((dynamic_quant_fp8_kernel<FP16Tag, BLOCK>), dim3(M),
                       dim3(BLOCK), 0, stream,
                       (uint8_t*)out.data_ptr(), scales.data_ptr<float>(),
                       (const short*)x.data_ptr(), K);
  }
  HIP_CHECK_KERNEL();
}

void scale_rows_cols(torch::Tensor y, torch::Tensor row_scales,
                     torch::Tensor col_scales,
                     c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(y.is_contiguous());
  TORCH_CHECK(y.scalar_type() == torch::kBFloat16,
              "fp8 GEMM output rescale expects bf16");
  const int M = y.size(0);
  const int N = y.size(1);
  TORCH_CHECK(N % 8 == 0);
  constexpr int BLOCK = 256;
  dim3 grid(ceil_div(N / 8, BLOCK), M);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const short* bptr =
      bias.has_value() ? (const short*)bias->data_ptr() : nullptr;
  hipLaunchKernelGGL((scale_rows_cols_kernel<BLOCK>), grid, dim3(BLOCK), 0,
                     stream, (short*)y.data_ptr(),
                     row_scales.data_ptr<float>(),
                     col_scales.data_ptr<float>(), bptr, N);
  HIP_CHECK_KERNEL();
}

}  // namespace vllm_amd
