// Fused rotary position embedding (neox/Llama style) for q and k, gfx950.
//
// cos/sin are host-precomputed (fp32 table [max_pos, D/2] each, packed as
// [max_pos, D] = cos|sin) per the CDNA4 guide's trig-table rule: on-device
// sinf/cosf turns a memory-bound op VALU-bound (guide Appendix B).
#include "common.h"

// q: [T, Hq*D], k: [T, Hkv*D]; rotate pairs (i, i+D/2) within each head.
__global__ void rope_kernel(ushort* __restrict__ q, ushort* __restrict__ k,
                            const float* __restrict__ cos_sin,
                            const int* __restrict__ positions,
                            int T, int Hq, int Hkv, int D) {
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
      base = q + ((long long)t * Hq + h) * D;
    } else {
      base = k + ((long long)t * Hkv + (h - Hq)) * D;
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
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous());
  int T = q.size(0);
  int D = cos_sin.size(1);
  TORCH_CHECK(q.numel() == (long long)T * num_q_heads * D, "rope: q shape");
  TORCH_CHECK(k.numel() == (long long)T * num_kv_heads * D, "rope: k shape");
  long long total = (long long)T * (num_q_heads + num_kv_heads) * (D / 2);
  long long blocks = (total + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rope_kernel, dim3((int)blocks), dim3(256), 0, stream,
                     (ushort*)q.data_ptr(), (ushort*)k.data_ptr(),
                     cos_sin.data_ptr<float>(), positions.data_ptr<int>(),
                     T, (int)num_q_heads, (int)num_kv_heads, D);
}
