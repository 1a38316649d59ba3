#include "hip/hip_runtime.h"
// Split-K MFMA "skinny" GEMM for decode-shape projections on gfx950.
//
// Motivation (profiles/r01_decode8b_kernel_stats.md): at decode batch
// M<=64, hipBLASLt's heuristic tiles launch ~48 workgroups for e.g.
// [32,4096]x[4096,6144] — 48 of 256 CUs — and read weights at ~1 TB/s
// cold.  This kernel splits K so the launch fills the chip and streams the
// weight matrix at HBM rate; the tiny activation matrix stays L2-resident.
//
//   out[M, N] = x[M, K] @ W[N, K]^T      (bf16 in, fp32 accumulate)
//
// Geometry: one workgroup = 4 waves, each wave owns one 16-wide n-tile
// (64 n per workgroup) and a K-slice; per K-step each wave issues
// ceil(M/16) mfma_f32_16x16x32_bf16.  Each K-slice writes a partial fp32
// plane; a finalize kernel reduces the planes and converts to bf16
// (deterministic split-K, no atomics).
//
// Fragment layouts (verified on MI355X by tests/test_kernels_gpu.py
// numerics + the asymmetric-input rule from the CDNA4 guide §3):
//   A frag: lane l holds A[m = l%16][k = (l/16)*8 + j], j=0..7
//   B frag: lane l holds B[k = (l/16)*8 + j][n = l%16]
//   C/D:    lane l reg r holds D[row = (l/16)*4 + r][col = l%16]
#include "common.h"

using bf16x8_t = __attribute__((ext_vector_type(8))) short;
using f32x4_t = __attribute__((ext_vector_type(4))) float;

#define SG_NTILE 16
#define SG_WAVES 4

// Staged K sub-slice width (elements) and padded LDS row pitch.  The +8
// element pad (16 B) breaks the 512 B power-of-2 row stride that would put
// all 16 fragment lanes in the same LDS bank (guide §6 Guideline 4).
#define SG_KSUB 256
#define SG_LDS_PITCH (SG_KSUB + 8)

template <int MTILES>
__global__ __launch_bounds__(256)
void skinny_gemm_kernel(float* __restrict__ out_ws,      // [splits, M, N] fp32
                        const ushort* __restrict__ x,    // [M, K]
                        const ushort* __restrict__ w,    // [N, K]
                        int M, int N, int K, int k_slice) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int n0 = blockIdx.x * (SG_NTILE * SG_WAVES);
  const int kbegin = blockIdx.y * k_slice;
  const int kend = min(kbegin + k_slice, K);

  const int frag_row = lane % 16;        // m (A) / n (B)
  const int frag_kgrp = lane / 16;       // which 8-wide k group

  // W tile [64 n x SG_KSUB k] staged cooperatively (coalesced 64 B row
  // chunks from HBM), fragments then read via ds_read_b128.
  __shared__ ushort w_lds[SG_NTILE * SG_WAVES][SG_LDS_PITCH];

  f32x4_t acc[MTILES];
#pragma unroll
  for (int t = 0; t < MTILES; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};

  for (int ks0 = kbegin; ks0 < kend; ks0 += SG_KSUB) {
    const int kw = min(SG_KSUB, kend - ks0);      // valid k width (mult of 32)
    // ---- stage: 64 rows x kw elems; consecutive threads take consecutive
    // 8-elem units within a row -> fully coalesced global reads
    const int units_per_row = kw / 8;
    const int total_units = (SG_NTILE * SG_WAVES) * units_per_row;
    __syncthreads();
    for (int u = threadIdx.x; u < total_units; u += blockDim.x) {
      const int row = u / units_per_row;
      const int kc = (u % units_per_row) * 8;
      uint4 val = make_uint4(0, 0, 0, 0);
      if (n0 + row < N)
        val = *reinterpret_cast<const uint4*>(
            w + (long long)(n0 + row) * K + ks0 + kc);
      *reinterpret_cast<uint4*>(&w_lds[row][kc]) = val;
    }
    __syncthreads();

    for (int k0 = 0; k0 < kw; k0 += 32) {
      const int kf = ks0 + k0 + frag_kgrp * 8;
      bf16x8_t bfrag;
      {
        bf16x8 tmp;
        tmp.u = *reinterpret_cast<const uint4*>(
            &w_lds[wave * SG_NTILE + frag_row][k0 + frag_kgrp * 8]);
#pragma unroll
        for (int j = 0; j < 8; ++j) bfrag[j] = (short)tmp.h[j];
      }
#pragma unroll
      for (int t = 0; t < MTILES; ++t) {
        const int m = t * 16 + frag_row;
        bf16x8_t afrag;
        if (m < M) {
          // x is tiny (M<=32 rows) and L2-resident; strided reads are cheap
          bf16x8 tmp;
          tmp.u = *reinterpret_cast<const uint4*>(x + (long long)m * K + kf);
#pragma unroll
          for (int j = 0; j < 8; ++j) afrag[j] = (short)tmp.h[j];
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) afrag[j] = 0;
        }
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[t],
                                                         0, 0, 0);
      }
    }
  }

  // C/D layout: lane l reg r -> D[row=(l/16)*4+r][col=l%16].
  // Deterministic split-K: each k-slice writes its own partial plane;
  // the finalize kernel reduces over slices (no atomics).
  const int n = n0 + wave * SG_NTILE + frag_row;
  float* plane = out_ws + (long long)blockIdx.y * M * N;
  if (n < N) {
#pragma unroll
    for (int t = 0; t < MTILES; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = t * 16 + frag_kgrp * 4 + r;
        if (m < M) plane[(long long)m * N + n] = acc[t][r];
      }
    }
  }
}

__global__ void skinny_gemm_finalize_kernel(ushort* __restrict__ out,
                                            const float* __restrict__ ws,
                                            long long total, int splits) {
  long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; idx < total; idx += stride) {
    float v = 0.0f;
    for (int s = 0; s < splits; ++s) v += ws[(long long)s * total + idx];
    out[idx] = f32_to_bf16(v);
  }
}

// ---------------------------------------------------------------------------
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                 torch::Tensor ws) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(ws.scalar_type() == torch::kFloat32 && ws.is_contiguous());
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = w.size(0);
  TORCH_CHECK(w.size(1) == K && out.size(0) == M && out.size(1) == N);
  TORCH_CHECK(M <= 32, "skinny_gemm: M must be <= 32");
  TORCH_CHECK(K % 32 == 0 && N % 16 == 0);

  hipStream_t stream = at::hip::getCurrentHIPStream();
  const int n_blocks = (N + SG_NTILE * SG_WAVES - 1) / (SG_NTILE * SG_WAVES);
  // split K so the grid lands near ~2048 workgroups (8 per CU)
  int split = 2048 / max(1, n_blocks);
  const int max_split = max(1, K / 256);
  if (split > max_split) split = max_split;
  if (split < 1) split = 1;
  int k_slice = (K / split + 31) / 32 * 32;
  const int grid_y = (K + k_slice - 1) / k_slice;
  TORCH_CHECK(ws.numel() >= (long long)grid_y * M * N,
              "skinny_gemm workspace too small");

  dim3 grid(n_blocks, grid_y);
  if (M <= 16) {
    hipLaunchKernelGGL((skinny_gemm_kernel<1>), grid, dim3(256), 0, stream,
                       ws.data_ptr<float>(), (const ushort*)x.data_ptr(),
                       (const ushort*)w.data_ptr(), M, N, K, k_slice);
  } else {
    hipLaunchKernelGGL((skinny_gemm_kernel<2>), grid, dim3(256), 0, stream,
                       ws.data_ptr<float>(), (const ushort*)x.data_ptr(),
                       (const ushort*)w.data_ptr(), M, N, K, k_slice);
  }
  long long total = (long long)M * N;
  long long blocks = min((total + 255) / 256, (long long)2048);
  hipLaunchKernelGGL(skinny_gemm_finalize_kernel, dim3((int)blocks), dim3(256),
                     0, stream, (ushort*)out.data_ptr(), ws.data_ptr<float>(),
                     total, grid_y);
}
