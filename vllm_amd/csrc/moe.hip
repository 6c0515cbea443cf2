// MoE grouped-GEMM kernels for CDNA4 (gfx950).
//
// Role of the reference's Triton grouped GEMM
// (vllm/model_executor/layers/fused_moe/fused_moe.py:299) and
// moe_align_block_size (csrc/libtorch_stable/moe/moe_align_sum_kernels.cu:326),
// redesigned as hand-written MFMA kernels: tokens are sorted by expert
// into BLOCK_M=64 tiles on-device (no host sync — shapes depend only on
// (T, topk, E), so the whole MoE layer is hipGraph-capturable), then a
// tiled 16x16x32-bf16 MFMA GEMM streams each expert's weights from HBM
// once per m-tile. Decode-shape MoE is weight-bound (~32 flops per
// weight byte at BM=64), so the kernel targets the HBM roofline, not
// peak MFMA.
//
// Pipeline (ops/hip_ops.py fused_moe):
//   moe_align: counts -> tile-aligned offsets -> sorted flat ids + the
//              inverse permutation (combine is deterministic, no atomics)
//   moe_gemm(hidden, w13)   -> y [EM, 2I]   (A rows gathered via sorted)
//   silu_and_mul(y)         -> act [EM, I]  (existing elementwise kernel)
//   moe_gemm(act, w2)       -> y2 [EM, H]   (A rows in sorted order)
//   moe_combine: out[t] = sum_k w[t,k] * y2[inv_perm[t*K+k]]

#include <torch/all.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "common.h"

namespace vllm_amd {

namespace moe {

constexpr int BM = 64;    // token rows per tile (must divide align padding)
constexpr int BN = 128;   // output cols per block
constexpr int BK = 64;    // K step staged in LDS
constexpr int THREADS = 256;
constexpr int SENTINEL = 0x7fffffff;

// --------------------------------------------------------------------------
// Alignment kernels.
__global__ void zero_kernel(int* counts, int* fill, int e) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < e) {
    counts[i] = 0;
    fill[i] = 0;
  }
}

__global__ void count_kernel(const int* __restrict__ topk_ids, int total,
                             int* __restrict__ counts) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < total) atomicAdd(&counts[topk_ids[i]], 1);
}

// One block: tile-aligned exclusive offsets, expert id per tile, sorted
// array pre-filled with SENTINEL. E <= a few hundred, EM up to ~1M.
__global__ void scan_kernel(const int* __restrict__ counts, int e,
                            int* __restrict__ off,
                            int* __restrict__ expert_tiles, int max_tiles,
                            int* __restrict__ sorted, int em_max) {
  __shared__ int total_tiles_sh;
  if (threadIdx.x == 0) {
    int tok_off = 0;
    for (int i = 0; i < e; ++i) {
      off[i] = tok_off;
      tok_off += (counts[i] + BM - 1) / BM * BM;
    }
    total_tiles_sh = tok_off / BM;
  }
  __syncthreads();
  const int total_tiles = total_tiles_sh;
  for (int t = threadIdx.x; t < max_tiles; t += blockDim.x) {
    int ex = -1;
    if (t < total_tiles) {
      // tile t belongs to the expert whose [off, off+aligned) covers it.
      // Linear scan per tile is fine: E small, one block total.
      for (int i = e - 1; i >= 0; --i) {
        if (t * BM >= off[i]) {
          ex = i;
          break;
        }
      }
    }
    expert_tiles[t] = ex;
  }
  for (int i = threadIdx.x; i < em_max; i += blockDim.x)
    sorted[i] = SENTINEL;
}

__global__ void scatter_kernel(const int* __restrict__ topk_ids, int total,
                               const int* __restrict__ off,
                               int* __restrict__ fill,
                               int* __restrict__ sorted,
                               int* __restrict__ inv_perm) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  int e = topk_ids[i];
  int pos = off[e] + atomicAdd(&fill[e], 1);
  sorted[pos] = i;
  inv_perm[i] = pos;
}

// --------------------------------------------------------------------------
// Grouped GEMM: C[m, n] = sum_k A[row(m), k] * B[expert(m), n, k].
// Per 16x16x32 MFMA: A-frag lane&15 = row, k = (lane>>4)*8 + j;
// B-frag lane&15 = col, same k split; C-frag col = lane&15,
// row = (lane>>4)*4 + reg (conventions proven in attention_prefill.hip).
//
// LDS XOR swizzle (guide T2): byte offset ^= (row&7)<<4 spreads the 16
// same-column rows of a fragment read across 8 distinct 16B slots.
DEVINL int swz(int row, int byte_off) { return byte_off ^ ((row & 7) << 4); }

template <typename Tag>
__launch_bounds__(THREADS) __global__
void moe_gemm_kernel(const short* __restrict__ A, long lda,
                     const short* __restrict__ B, long expert_stride,
                     long ldb, short* __restrict__ C, long ldc,
                     const int* __restrict__ sorted,
                     const int* __restrict__ expert_tiles, int K, int N,
                     int topk_div, int total_flat) {
  const int tile_m = blockIdx.y;
  const int n0 = blockIdx.x * BN;
  const int expert = expert_tiles[tile_m];
  if (expert < 0) return;  // padding tile

  __shared__ short a_lds[BM * BK];  // [row][k], swizzled rows
  __shared__ short b_lds[BN * BK];  // [col][k], swizzled cols
  __shared__ int rows_sh[BM];       // A source row per tile row (-1 pad)
  __shared__ int cval_sh[BM];       // sorted id per tile row (C row valid)

  const int tid = threadIdx.x;
  if (tid < BM) {
    int sid = sorted[tile_m * BM + tid];
    cval_sh[tid] = (sid < total_flat) ? 1 : 0;
    rows_sh[tid] = (sid < total_flat)
                       ? (topk_div > 0 ? sid / topk_div : tile_m * BM + tid)
                       : -1;
  }
  __syncthreads();

  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int l16 = lane & 15;
  const int l4 = lane >> 4;       // 0..3: k-quarter for A/B frags
  const int wm = wave & 1;        // m 32-half (2 m-frags of 16)
  const int wn = wave >> 1;       // n 64-half (4 n-frags of 16)

  f32x4 acc[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[m][i] = f32x4{0.f, 0.f, 0.f, 0.f};

  const short* b_exp = B + (long)expert * expert_stride;

  for (int k0 = 0; k0 < K; k0 += BK) {
    // Stage A: BM rows x 64 cols bf16 = BM x 128B; 16B per thread-chunk.
    {
#pragma unroll
      for (int it = 0; it < BM * 8 / THREADS; ++it) {
        const int idx = it * THREADS + tid;
        const int r = idx >> 3;
        const int cb = (idx & 7) * 16;   // byte col 0..112
        const int arow = rows_sh[r];
        s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (arow >= 0)
          v = *reinterpret_cast<const s16x8*>(A + (long)arow * lda + k0 +
                                              (cb >> 1));
        *reinterpret_cast<s16x8*>(
            reinterpret_cast<char*>(a_lds) + r * 128 + swz(r, cb)) = v;
      }
    }
    // Stage B: 128 rows (n) x 64 cols = 128 x 128B; 4 x 16B per thread.
    {
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        const int idx = it * THREADS + tid;   // 0..1023
        const int n = idx >> 3;
        const int cb = (idx & 7) * 16;
        s16x8 v = *reinterpret_cast<const s16x8*>(
            b_exp + (long)(n0 + n) * ldb + k0 + (cb >> 1));
        *reinterpret_cast<s16x8*>(
            reinterpret_cast<char*>(b_lds) + n * 128 + swz(n, cb)) = v;
      }
    }
    __syncthreads();
    // 2 K-steps of 32 within the tile.
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int kb = kk * 64 + l4 * 16;  // byte offset of this lane's 8 elems
      s16x8 afrag[2];
#pragma unroll
      for (int m = 0; m < 2; ++m) {
        const int arow = wm * 32 + m * 16 + l16;
        afrag[m] = *reinterpret_cast<const s16x8*>(
            reinterpret_cast<const char*>(a_lds) + arow * 128 +
            swz(arow, kb));
      }
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int bcol = wn * 64 + nf * 16 + l16;
        s16x8 bfrag = *reinterpret_cast<const s16x8*>(
            reinterpret_cast<const char*>(b_lds) + bcol * 128 +
            swz(bcol, kb));
#pragma unroll
        for (int m = 0; m < 2; ++m)
          acc[m][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[m], bfrag, acc[m][nf], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // Epilogue: C rows are sorted positions; skip padding rows.
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int n = n0 + wn * 64 + nf * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int mrow = wm * 32 + m * 16 + l4 * 4 + r;
        if (cval_sh[mrow])
          C[(long)(tile_m * BM + mrow) * ldc + n] =
              from_f32<Tag>(acc[m][nf][r]);
      }
    }
  }
}

// --------------------------------------------------------------------------
// Fragment-major grouped GEMM: B is pre-shuffled at weight-load time
// into MFMA-fragment order [E][N/16][K/32][64 lanes][8 elems] so each
// wave reads its B-fragment as ONE coalesced 16B/lane chunk straight
// from HBM. A-fragments load per-lane from global (scattered but tiny —
// <1% of traffic, L3-resident across the N/128 blocks sharing the
// m-tile). ZERO LDS and ZERO barriers in the k-loop: nothing forces a
// vmcnt(0) drain, so the compiler pipelines the weight stream at full
// depth (the staged-LDS variant stalls on the barrier drain — guide
// §Composed models). This is the MI355X-first inference layout:
// weights are written once and streamed billions of times, so they
// belong in consumer order.
template <typename Tag>
__launch_bounds__(THREADS) __global__
void moe_gemm_shuf_kernel(const short* __restrict__ A, long lda,
                          const short* __restrict__ B, long expert_stride,
                          short* __restrict__ C, long ldc,
                          const int* __restrict__ sorted,
                          const int* __restrict__ expert_tiles, int K,
                          int N, int topk_div, int total_flat) {
  const int tile_m = blockIdx.y;
  const int n0 = blockIdx.x * BN;
  const int expert = expert_tiles[tile_m];
  if (expert < 0) return;  // padding tile

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int l16 = lane & 15;
  const int l4 = lane >> 4;
  const int wm = wave & 1;   // m 32-half (2 m-frags)
  const int wn = wave >> 1;  // n 64-half (4 n-frags)

  // Per-lane A row pointers (A-frag: lane&15 = row, k = (lane>>4)*8+j).
  // Padding rows read row 0 (real data); their C rows are masked at the
  // epilogue and never mix into valid rows (MFMA rows are independent).
  const short* a_ptr[2];
  bool cval[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m) {
    const int r = tile_m * BM + wm * 32 + m * 16 + l16;
    const int sid = sorted[r];
    const long arow =
        (sid < total_flat)
            ? (topk_div > 0 ? (long)(sid / topk_div) : (long)r)
            : 0;
    a_ptr[m] = A + arow * lda + l4 * 8;
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int crow = tile_m * BM + wm * 32 + m * 16 + l4 * 4 + rr;
      cval[m][rr] = sorted[crow] < total_flat;
    }
  }

  f32x4 acc[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[m][i] = f32x4{0.f, 0.f, 0.f, 0.f};

  // B fragment stream for this wave: frags at n-tiles n0/16 + wn*4 + nf,
  // granule 64 lanes x 8 elems; consecutive k-steps are contiguous 1 KB
  // lines, so each wave walks 4 linear HBM streams. A re-reads hit the
  // L3 (the m-tile's rows are shared by all N/128 blocks).
  const int ksteps = K / 32;
  const short* bptr[4];
#pragma unroll
  for (int nf = 0; nf < 4; ++nf)
    bptr[nf] = B + (long)expert * expert_stride +
               ((long)((n0 >> 4) + wn * 4 + nf) * ksteps) * 512 + lane * 8;

  // No LDS, no barriers: the compiler software-pipelines the 6 loads +
  // 8 MFMAs per k-step with fine-grained s_waitcnt.
  for (int ks = 0; ks < ksteps; ++ks) {
    s16x8 afrag[2];
#pragma unroll
    for (int m = 0; m < 2; ++m)
      afrag[m] = *reinterpret_cast<const s16x8*>(a_ptr[m] + ks * 32);
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const s16x8 bfrag =
          *reinterpret_cast<const s16x8*>(bptr[nf] + (long)ks * 512);
#pragma unroll
      for (int m = 0; m < 2; ++m)
        acc[m][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[m], bfrag, acc[m][nf], 0, 0, 0);
    }
  }

#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int n = n0 + wn * 64 + nf * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        if (cval[m][r])
          C[(long)(tile_m * BM + wm * 32 + m * 16 + l4 * 4 + r) * ldc + n] =
              from_f32<Tag>(acc[m][nf][r]);
      }
    }
  }
}

// --------------------------------------------------------------------------
// Weighted combine: out[t, :] = sum_k w[t*K+k] * Y[inv_perm[t*K+k], :].
template <typename Tag>
__global__ void combine_kernel(short* __restrict__ out,
                               const short* __restrict__ Y, long ldy,
                               const float* __restrict__ weights,
                               const int* __restrict__ inv_perm, int topk,
                               int H) {
  const int t = blockIdx.x;
  for (int h8 = threadIdx.x; h8 * 8 < H; h8 += blockDim.x) {
    float acc[8] = {0.f};
    for (int k = 0; k < topk; ++k) {
      const float w = weights[t * topk + k];
      const long row = inv_perm[t * topk + k];
      s16x8 v = *reinterpret_cast<const s16x8*>(Y + row * ldy + h8 * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += w * to_f32<Tag>(v[j]);
    }
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = from_f32<Tag>(acc[j]);
    *reinterpret_cast<s16x8*>(out + (long)t * H + h8 * 8) = o;
  }
}

}  // namespace moe

// Host-side entry points -----------------------------------------------------

void moe_align(torch::Tensor topk_ids, int64_t num_experts,
               torch::Tensor sorted, torch::Tensor expert_tiles,
               torch::Tensor inv_perm, torch::Tensor counts,
               torch::Tensor fill, torch::Tensor off) {
  using namespace moe;
  TORCH_CHECK(topk_ids.scalar_type() == torch::kInt32 &&
              topk_ids.is_contiguous());
  const int total = (int)topk_ids.numel();
  const int e = (int)num_experts;
  const int em_max = (int)sorted.numel();
  const int max_tiles = (int)expert_tiles.numel();
  TORCH_CHECK(counts.numel() >= e && fill.numel() >= e && off.numel() >= e);
  TORCH_CHECK(inv_perm.numel() >= total);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int* ids = topk_ids.data_ptr<int>();
  hipLaunchKernelGGL(zero_kernel, dim3(ceil_div(e, 256)), dim3(256), 0,
                     stream, counts.data_ptr<int>(), fill.data_ptr<int>(), e);
  hipLaunchKernelGGL(count_kernel, dim3(ceil_div(total, 256)), dim3(256), 0,
                     stream, ids, total, counts.data_ptr<int>());
  hipLaunchKernelGGL(scan_kernel, dim3(1), dim3(256), 0, stream,
                     counts.data_ptr<int>(), e, off.data_ptr<int>(),
                     expert_tiles.data_ptr<int>(), max_tiles,
                     sorted.data_ptr<int>(), em_max);
  hipLaunchKernelGGL(scatter_kernel, dim3(ceil_div(total, 256)), dim3(256),
                     0, stream, ids, total, off.data_ptr<int>(),
                     fill.data_ptr<int>(), sorted.data_ptr<int>(),
                     inv_perm.data_ptr<int>());
  HIP_CHECK_KERNEL();
}

void moe_gemm(torch::Tensor a, torch::Tensor b, torch::Tensor c,
              torch::Tensor sorted, torch::Tensor expert_tiles,
              int64_t topk_div, int64_t total_flat) {
  using namespace moe;
  TORCH_CHECK(a.dim() == 2 && b.dim() == 3 && c.dim() == 2);
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous() && c.is_contiguous());
  TORCH_CHECK(a.scalar_type() == b.scalar_type() &&
              a.scalar_type() == c.scalar_type());
  const int K = (int)a.size(1);
  const int N = (int)b.size(1);
  TORCH_CHECK(b.size(2) == K, "B inner dim mismatch");
  TORCH_CHECK(K % BK == 0, "K must be a multiple of ", BK);
  TORCH_CHECK(N % BN == 0, "N must be a multiple of ", BN);
  const int em = (int)c.size(0);
  TORCH_CHECK(em % BM == 0 && em / BM <= (int)expert_tiles.numel());
  TORCH_CHECK(c.size(1) == N);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid(N / BN, em / BM);
  auto launch = [&](auto tag) {
    using Tag = decltype(tag);
    hipLaunchKernelGGL(
        (moe_gemm_kernel<Tag>), grid, dim3(THREADS), 0, stream,
        (const short*)a.data_ptr(), a.size(1), (const short*)b.data_ptr(),
        b.size(1) * b.size(2), b.size(2), (short*)c.data_ptr(), c.size(1),
        sorted.data_ptr<int>(), expert_tiles.data_ptr<int>(), K, N,
        (int)topk_div, (int)total_flat);
  };
  if (a.scalar_type() == torch::kBFloat16)
    launch(BF16Tag{});
  else if (a.scalar_type() == torch::kHalf)
    launch(FP16Tag{});
  else
    TORCH_CHECK(false, "moe_gemm: bf16/fp16 only");
  HIP_CHECK_KERNEL();
}

void moe_combine(torch::Tensor out, torch::Tensor y,
                 torch::Tensor topk_weights, torch::Tensor inv_perm) {
  using namespace moe;
  TORCH_CHECK(out.dim() == 2 && y.dim() == 2);
  TORCH_CHECK(out.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(topk_weights.scalar_type() == torch::kFloat);
  const int T = (int)out.size(0);
  const int H = (int)out.size(1);
  const int topk = (int)topk_weights.size(1);
  TORCH_CHECK(H % 8 == 0);
  TORCH_CHECK(y.size(1) == H);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  auto launch = [&](auto tag) {
    using Tag = decltype(tag);
    hipLaunchKernelGGL((combine_kernel<Tag>), dim3(T),
                       dim3(std::min(256, std::max(64, H / 8))), 0, stream,
                       (short*)out.data_ptr(), (const short*)y.data_ptr(),
                       y.size(1), topk_weights.data_ptr<float>(),
                       inv_perm.data_ptr<int>(), topk, H);
  };
  if (out.scalar_type() == torch::kBFloat16)
    launch(BF16Tag{});
  else if (out.scalar_type() == torch::kHalf)
    launch(FP16Tag{});
  else
    TORCH_CHECK(false, "moe_combine: bf16/fp16 only");
  HIP_CHECK_KERNEL();
}


void moe_gemm_shuf(torch::Tensor a, torch::Tensor b_shuf, torch::Tensor c,
                   torch::Tensor sorted, torch::Tensor expert_tiles,
                   int64_t n, int64_t k, int64_t topk_div,
                   int64_t total_flat) {
  using namespace moe;
  TORCH_CHECK(a.dim() == 2 && c.dim() == 2);
  TORCH_CHECK(a.is_contiguous() && b_shuf.is_contiguous() &&
              c.is_contiguous());
  TORCH_CHECK(a.scalar_type() == b_shuf.scalar_type() &&
              a.scalar_type() == c.scalar_type());
  const int K = (int)k;
  const int N = (int)n;
  TORCH_CHECK(a.size(1) == K);
  TORCH_CHECK(K % 64 == 0 && N % BN == 0);
  const int em = (int)c.size(0);
  TORCH_CHECK(em % BM == 0 && em / BM <= (int)expert_tiles.numel());
  TORCH_CHECK(c.size(1) == N);
  const long expert_stride = (long)(N / 16) * (K / 32) * 512;
  TORCH_CHECK(b_shuf.numel() % expert_stride == 0);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid(N / BN, em / BM);
  auto launch = [&](auto tag) {
    using Tag = decltype(tag);
    hipLaunchKernelGGL(
        (moe_gemm_shuf_kernel<Tag>), grid, dim3(THREADS), 0, stream,
        (const short*)a.data_ptr(), a.size(1),
        (const short*)b_shuf.data_ptr(), expert_stride,
        (short*)c.data_ptr(), c.size(1), sorted.data_ptr<int>(),
        expert_tiles.data_ptr<int>(), K, N, (int)topk_div,
        (int)total_flat);
  };
  if (a.scalar_type() == torch::kBFloat16)
    launch(BF16Tag{});
  else if (a.scalar_type() == torch::kHalf)
    launch(FP16Tag{});
  else
    TORCH_CHECK(false, "moe_gemm_shuf: bf16/fp16 only");
  HIP_CHECK_KERNEL();
}

}  // namespace vllm_amd
