// RMSNorm (+fused residual add) and SiLU-mul kernels for gfx950.
//
// These are the memory-bound elementwise/normalization hot ops of the
// lws_amd served engine (SURVEY.md §2.10 kernel inventory).  Both are
// written for the HBM roofline: bf16 I/O vectorized as 16 B/lane, fp32
// accumulation, one pass over the data, grid-stride over rows so the
// launch fills 256 CUs without oversubscribing the scheduler.
#include "common.h"

// ---------------------------------------------------------------------------
// rmsnorm: out[r, :] = x[r, :] * rsqrt(mean(x^2) + eps) * w
// One workgroup per row (grid-stride).  D must be a multiple of 8.
__global__ void rmsnorm_kernel(ushort* __restrict__ out,
                               const ushort* __restrict__ in,
                               const ushort* __restrict__ weight,
                               float eps, int rows, int D) {
  __shared__ float lds[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const ushort* x = in + (size_t)row * D;
    ushort* o = out + (size_t)row * D;
    float ss = 0.0f;
    for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
      bf16x8 v;
      v.u = *reinterpret_cast<const uint4*>(x + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_to_f32(v.h[j]);
        ss += f * f;
      }
    }
    ss = block_reduce_sum(ss, lds);
    float inv = rsqrtf(ss / (float)D + eps);
    for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
      bf16x8 v, w, r;
      v.u = *reinterpret_cast<const uint4*>(x + i);
      w.u = *reinterpret_cast<const uint4*>(weight + i);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        r.h[j] = f32_to_bf16(bf16_to_f32(v.h[j]) * inv * bf16_to_f32(w.h[j]));
      *reinterpret_cast<uint4*>(o + i) = r.u;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// fused_add_rmsnorm: residual[r,:] += x[r,:]; x[r,:] = norm(residual[r,:])*w
// (in-place contract of the serving engine's per-layer residual stream;
// fusing the add saves one full HBM round-trip per layer)
__global__ void fused_add_rmsnorm_kernel(ushort* __restrict__ x,
                                         ushort* __restrict__ residual,
                                         const ushort* __restrict__ weight,
                                         float eps, int rows, int D) {
  __shared__ float lds[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    ushort* xr = x + (size_t)row * D;
    ushort* rr = residual + (size_t)row * D;
    float ss = 0.0f;
    for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
      bf16x8 a, b, s;
      a.u = *reinterpret_cast<const uint4*>(xr + i);
      b.u = *reinterpret_cast<const uint4*>(rr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_to_f32(a.h[j]) + bf16_to_f32(b.h[j]);
        s.h[j] = f32_to_bf16(f);
        // accumulate over the bf16-rounded sum: matches the two-op
        // reference (residual = residual + x; norm(residual)) numerics
        float fr = bf16_to_f32(s.h[j]);
        ss += fr * fr;
      }
      *reinterpret_cast<uint4*>(rr + i) = s.u;
    }
    ss = block_reduce_sum(ss, lds);
    float inv = rsqrtf(ss / (float)D + eps);
    for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
      bf16x8 s, w, o;
      s.u = *reinterpret_cast<const uint4*>(rr + i);
      w.u = *reinterpret_cast<const uint4*>(weight + i);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.h[j] = f32_to_bf16(bf16_to_f32(s.h[j]) * inv * bf16_to_f32(w.h[j]));
      *reinterpret_cast<uint4*>(xr + i) = o.u;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// silu_mul: out[r, i] = silu(gu[r, i]) * gu[r, I + i]   (gate-up fused proj)
__global__ void silu_mul_kernel(ushort* __restrict__ out,
                                const ushort* __restrict__ gateup,
                                long long total, int I) {
  long long idx = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (; idx < total; idx += stride) {
    long long row = idx / I;
    long long col = idx % I;
    const ushort* g = gateup + row * (2LL * I) + col;
    const ushort* u = g + I;
    bf16x8 gv, uv, ov;
    gv.u = *reinterpret_cast<const uint4*>(g);
    uv.u = *reinterpret_cast<const uint4*>(u);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32(gv.h[j]);
      float uf = bf16_to_f32(uv.h[j]);
      float s = gf / (1.0f + __expf(-gf));
      ov.h[j] = f32_to_bf16(s * uf);
    }
    *reinterpret_cast<uint4*>(out + row * I + col) = ov.u;
  }
}

// ---------------------------------------------------------------------------
// host launchers (called from bindings.cpp)
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

static inline int norm_grid(int rows) {
  // cap at ~8 blocks/CU x 256 CUs (Guideline 11), grid-stride the rest
  int cap = 2048;
  return rows < cap ? rows : cap;
}

void rmsnorm(torch::Tensor out, torch::Tensor input, torch::Tensor weight,
             double eps) {
  TORCH_CHECK(input.is_cuda() && input.scalar_type() == torch::kBFloat16,
              "rmsnorm: bf16 GPU tensors required");
  TORCH_CHECK(input.is_contiguous() && out.is_contiguous());
  int D = input.size(-1);
  TORCH_CHECK(D % 8 == 0, "rmsnorm: hidden dim must be a multiple of 8");
  int rows = input.numel() / D;
  dim3 block(256);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsnorm_kernel, dim3(norm_grid(rows)), block, 0, stream,
                     (ushort*)out.data_ptr(), (const ushort*)input.data_ptr(),
                     (const ushort*)weight.data_ptr(), (float)eps, rows, D);
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor weight, double eps) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && residual.is_contiguous());
  int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0);
  int rows = x.numel() / D;
  dim3 block(256);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(fused_add_rmsnorm_kernel, dim3(norm_grid(rows)), block, 0,
                     stream, (ushort*)x.data_ptr(), (ushort*)residual.data_ptr(),
                     (const ushort*)weight.data_ptr(), (float)eps, rows, D);
}

void silu_mul(torch::Tensor out, torch::Tensor gateup) {
  TORCH_CHECK(gateup.is_cuda() && gateup.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(gateup.is_contiguous() && out.is_contiguous());
  int I = gateup.size(-1) / 2;
  TORCH_CHECK(I % 8 == 0, "silu_mul: intermediate dim must be a multiple of 8");
  long long total = (long long)(gateup.numel() / (2LL * I)) * I;
  dim3 block(256);
  long long blocks = (total / 8 + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(silu_mul_kernel, dim3((int)blocks), block, 0, stream,
                     (ushort*)out.data_ptr(), (const ushort*)gateup.data_ptr(),
                     total, I);
}
