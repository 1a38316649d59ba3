// Fused rotary position embedding (neox/Llama style) for q and k, gfx950.
//
// cos/sin are host-precomputed (fp32 table [max_pos, D] = cos|sin halves)
// per the CDNA4 guide's trig-table rule: on-device sinf/cosf turns a
// memory-bound op VALU-bound (guide Appendix B).  q and k may be strided
// row views straight into the fused qkv GEMM output (zero-copy decode).
#include "common.h"

// q: [T, Hq, D] rows at q_stride elems; k: [T, Hkv, D] rows at k_stride.
__global__ void rope_kernel(ushort* __restrict__ q, ushort* __restrict__ k,
                            const float* __restrict__ cos_sin,
                            const int* __restrict__ positions,
                            int T, int Hq, int Hkv, int D,
                            long long q_stride, long long k_stride) {
  const int half = D / 2;
  const long long total = (long long)T * (Hq + Hkv) * half;
  long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; idx < total; idx += stride) {
    const int i = (int)(idx % half);
    long long rest = idx / half;
    const int h = (int)(rest % (Hq + Hkv));
    const int t = (int)(rest / (Hq + Hkv));
    const int pos = positions[t];
    const float c = cos_sin[(long long)pos * D + i];
    const float s = cos_sin[(long long)pos * D + half + i];
    ushort* base;
    if (h < Hq) {
      base = q + (long long)t * q_stride + (long long)h * D;
    } else {
      base = k + (long long)t * k_stride + (long long)(h - Hq) * D;
    }
    float x1 = bf16_to_f32(base[i]);
    float x2 = bf16_to_f32(base[i + half]);
    base[i] = f32_to_bf16(x1 * c - x2 * s);
    base[i + half] = f32_to_bf16(x2 * c + x1 * s);
  }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void rope(torch::Tensor q, torch::Tensor k, torch::Tensor cos_sin,
          torch::Tensor positions, long long num_q_heads, long long num_kv_heads) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32);
  TORCH_CHECK(positions.scalar_type() == torch::kInt32);
  int T = q.size(0);
  int D = cos_sin.size(1);
  // accept [T, H*D] (stride(1)==1) or [T, H, D] views with contiguous rows
  long long q_stride, k_stride;
  if (q.dim() == 2) {
    TORCH_CHECK(q.stride(1) == 1 && k.stride(1) == 1);
    q_stride = q.stride(0);
    k_stride = k.stride(0);
  } else {
    TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == D);
    TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == D);
    q_stride = q.stride(0);
    k_stride = k.stride(0);
  }
  long long total = (long long)T * (num_q_heads + num_kv_heads) * (D / 2);
  long long blocks = (total + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rope_kernel, dim3((int)blocks), dim3(256), 0, stream,
                     (ushort*)q.data_ptr(), (ushort*)k.data_ptr(),
                     cos_sin.data_ptr<float>(), positions.data_ptr<int>(),
                     T, (int)num_q_heads, (int)num_kv_heads, D,
                     q_stride, k_stride);
}
