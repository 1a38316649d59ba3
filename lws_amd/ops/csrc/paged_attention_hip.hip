#include "hip/hip_runtime.h"
// Paged-attention decode for gfx950 (MI355X), GQA-aware, flash-decoding
// sequence split.
//
// Layout (one KV read amortized over the whole GQA group — the right CDNA4
// decode design for Llama-70B TP shards where H_kv per GPU is small):
//   grid = (num_seqs, H_kv, seq_chunks)  x  block 256 threads (4 waves)
//   Each workgroup processes CHUNK keys of one sequence for one kv head and
//   ALL G q-heads of that kv head's GQA group (G <= 16).
//   Thread (key, slice) layout: lane = key_local*16 + slice; each lane loads
//   16 B (8 bf16) of the key row -> fully-coalesced 256 B per key row, 16
//   keys in flight per pass over the workgroup (64 keys per chunk pass with
//   4 waves).  Scores reduce within 16-lane groups via shfl; online softmax
//   per chunk; V accumulated in registers per (key, slice) and reduced
//   across keys at the end.
//   Chunk partials (m, l, acc) land in an fp32 workspace; a second kernel
//   reduces chunks (flash-decoding), so 256 CUs stay busy even at B=1.
//
// KV cache layout: [num_pages, H_kv, page_size, D] bf16, D = 128, page 16.
#include "common.h"

#define PA_HEAD_DIM 128
#define PA_SLICES 16                // 16 lanes x 8 bf16 = 128 elements
#define PA_KEYS_PER_PASS 16         // 256 threads / 16 slices
#define PA_MAX_GQA 16

__global__ __launch_bounds__(256)
void paged_attention_chunk_kernel(
    float* __restrict__ ws_acc,      // [B, Hkv, chunks, G, 128]
    float* __restrict__ ws_ml,       // [B, Hkv, chunks, G, 2]  (m, l)
    const ushort* __restrict__ q,    // [B, Hq, 128]
    const ushort* __restrict__ k_cache,  // [pages, Hkv, page, 128]
    const ushort* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [B, max_pages]
    const int* __restrict__ seq_lens,      // [B]
    float scale, int G, int Hkv, int page_size, int max_pages,
    int chunk_keys, int num_chunks) {
  const int b = blockIdx.x;
  const int hkv = blockIdx.y;
  const int chunk = blockIdx.z;
  const int seq_len = seq_lens[b];
  const int kstart = chunk * chunk_keys;
  if (kstart >= seq_len && chunk > 0) {
    // out-of-range chunk: mark empty partial
    if (threadIdx.x < G) {
      const long long mlbase =
          ((((long long)b * Hkv + hkv) * num_chunks + chunk) * G + threadIdx.x) * 2;
      ws_ml[mlbase] = -INFINITY;
      ws_ml[mlbase + 1] = 0.0f;
    }
    return;
  }
  const int kend = min(kstart + chunk_keys, seq_len);

  const int lane = threadIdx.x % WAVE_SIZE;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int slice = threadIdx.x % PA_SLICES;        // which 8-elem slice
  const int key_local = threadIdx.x / PA_SLICES;    // 0..15

  // q for the whole GQA group staged in LDS: [G][128] fp32 (pre-scaled)
  __shared__ float q_lds[PA_MAX_GQA][PA_HEAD_DIM];
  __shared__ float score_lds[PA_MAX_GQA][PA_KEYS_PER_PASS];

  for (int g = 0; g < G; ++g) {
    const ushort* qrow = q + (((long long)b * Hkv + hkv) * G + g) * PA_HEAD_DIM;
    for (int i = threadIdx.x * 8; i < PA_HEAD_DIM; i += blockDim.x * 8) {
      bf16x8 v;
      v.u = *reinterpret_cast<const uint4*>(qrow + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) q_lds[g][i + j] = bf16_to_f32(v.h[j]) * scale;
    }
  }
  __syncthreads();

  // online-softmax state per q-head (uniform across the workgroup)
  float m_run[PA_MAX_GQA], l_run[PA_MAX_GQA];
  // per-thread V accumulator: this thread's (key stream, slice) partial
  float acc[PA_MAX_GQA][8];
  for (int g = 0; g < G; ++g) {
    m_run[g] = -INFINITY;
    l_run[g] = 0.0f;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[g][j] = 0.0f;
  }

  for (int k0 = kstart; k0 < kend; k0 += PA_KEYS_PER_PASS) {
    const int key = k0 + key_local;
    const bool valid = key < kend;
    // locate the key's page
    bf16x8 kv;
    kv.u = make_uint4(0, 0, 0, 0);
    const ushort* vrow = nullptr;
    if (valid) {
      const int page = block_tables[(long long)b * max_pages + key / page_size];
      const int off = key % page_size;
      const ushort* krow = k_cache +
          ((((long long)page * Hkv + hkv) * page_size + off) * PA_HEAD_DIM);
      vrow = v_cache +
          ((((long long)page * Hkv + hkv) * page_size + off) * PA_HEAD_DIM);
      kv.u = *reinterpret_cast<const uint4*>(krow + slice * 8);
    }
    float kf[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) kf[j] = bf16_to_f32(kv.h[j]);

    // scores for all G heads: partial dot over this lane's 8 elems,
    // reduced across the 16 lanes of the key group
    for (int g = 0; g < G; ++g) {
      float p = 0.0f;
#pragma unroll
      for (int j = 0; j < 8; ++j) p += kf[j] * q_lds[g][slice * 8 + j];
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) p += __shfl_xor(p, off, WAVE_SIZE);
      // lane slice==0 of each key group holds the full dot
      if (slice == 0) score_lds[g][key_local] = valid ? p : -INFINITY;
    }
    __syncthreads();

    // softmax update (uniform): each thread reads the 16 scores
    float pexp[PA_MAX_GQA];
    for (int g = 0; g < G; ++g) {
      float m_new = m_run[g];
#pragma unroll
      for (int i = 0; i < PA_KEYS_PER_PASS; ++i)
        m_new = fmaxf(m_new, score_lds[g][i]);
      float rescale = (m_run[g] == -INFINITY) ? 0.0f : __expf(m_run[g] - m_new);
      float s = score_lds[g][key_local];
      float e = (valid && s != -INFINITY) ? __expf(s - m_new) : 0.0f;
      pexp[g] = e;
      float lsum = 0.0f;
#pragma unroll
      for (int i = 0; i < PA_KEYS_PER_PASS; ++i) {
        float si = score_lds[g][i];
        lsum += (si == -INFINITY) ? 0.0f : __expf(si - m_new);
      }
      l_run[g] = l_run[g] * rescale + lsum;
      m_run[g] = m_new;
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[g][j] *= rescale;
    }
    __syncthreads();

    // V accumulate: this thread's key contributes p * v[slice]
    if (valid) {
      bf16x8 vv;
      vv.u = *reinterpret_cast<const uint4*>(vrow + slice * 8);
      float vf[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) vf[j] = bf16_to_f32(vv.h[j]);
      for (int g = 0; g < G; ++g) {
        const float p = pexp[g];
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[g][j] += p * vf[j];
      }
    }
  }

  // reduce acc over keys.  Within a wave, lanes with the same slice and
  // different key_local differ in lane bits 4,5 -> shfl_xor(16)+shfl_xor(32)
  // sums the wave's 4 key streams; every lane then holds its wave's slice
  // partial.  Cross-wave partials reduce through LDS.
  const long long wsbase =
      (((long long)b * Hkv + hkv) * num_chunks + chunk) * G * PA_HEAD_DIM;
  __shared__ float xwave[4][PA_SLICES][8];
  for (int g = 0; g < G; ++g) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = acc[g][j];
      v += __shfl_xor(v, 16, WAVE_SIZE);
      v += __shfl_xor(v, 32, WAVE_SIZE);
      acc[g][j] = v;
    }
    if (lane < PA_SLICES) {
#pragma unroll
      for (int j = 0; j < 8; ++j) xwave[wave][slice][j] = acc[g][j];
    }
    __syncthreads();
    // threads 0..127: (slice, j) sums over 4 waves
    if (threadIdx.x < PA_SLICES * 8) {
      const int s = threadIdx.x / 8;
      const int j = threadIdx.x % 8;
      float v = xwave[0][s][j] + xwave[1][s][j] + xwave[2][s][j] + xwave[3][s][j];
      ws_acc[wsbase + (long long)g * PA_HEAD_DIM + s * 8 + j] = v;
    }
    __syncthreads();
  }
  if (threadIdx.x < G) {
    const long long mlbase =
        ((((long long)b * Hkv + hkv) * num_chunks + chunk) * G + threadIdx.x) * 2;
    ws_ml[mlbase] = m_run[threadIdx.x];
    ws_ml[mlbase + 1] = l_run[threadIdx.x];
  }
}

// ---------------------------------------------------------------------------
// chunk reduction: combine per-chunk (m, l, acc) into the final output.
// grid = (B, Hq); block = 128 threads (one per output dim).
__global__ __launch_bounds__(128)
void paged_attention_reduce_kernel(
    ushort* __restrict__ out,        // [B, Hq, 128]
    const float* __restrict__ ws_acc,
    const float* __restrict__ ws_ml,
    const int* __restrict__ seq_lens,
    int G, int Hkv, int chunk_keys, int num_chunks) {
  const int b = blockIdx.x;
  const int hq = blockIdx.y;
  const int hkv = hq / G;
  const int g = hq % G;
  const int Hq = Hkv * G;
  const int used = min(num_chunks,
                       (seq_lens[b] + chunk_keys - 1) / chunk_keys);
  // global max
  float m_glob = -INFINITY;
  for (int c = 0; c < used; ++c) {
    const long long mlbase =
        ((((long long)b * Hkv + hkv) * num_chunks + c) * G + g) * 2;
    m_glob = fmaxf(m_glob, ws_ml[mlbase]);
  }
  float l_glob = 0.0f;
  float acc = 0.0f;
  const int d = threadIdx.x;
  for (int c = 0; c < used; ++c) {
    const long long mlbase =
        ((((long long)b * Hkv + hkv) * num_chunks + c) * G + g) * 2;
    const float m = ws_ml[mlbase];
    const float l = ws_ml[mlbase + 1];
    if (m == -INFINITY) continue;
    const float w = __expf(m - m_glob);
    l_glob += l * w;
    const long long abase =
        ((((long long)b * Hkv + hkv) * num_chunks + c) * G + g) * PA_HEAD_DIM;
    acc += ws_acc[abase + d] * w;
  }
  out[((long long)b * Hq + hq) * PA_HEAD_DIM + d] =
      f32_to_bf16(acc / fmaxf(l_glob, 1e-20f));
}

// ---------------------------------------------------------------------------
// reshape_and_cache: scatter new k/v token rows into the paged cache.
// k,v: [T, Hkv, 128]; slot_mapping: [T] (page*page_size + offset)
__global__ void reshape_and_cache_kernel(
    const ushort* __restrict__ k, const ushort* __restrict__ v,
    ushort* __restrict__ k_cache, ushort* __restrict__ v_cache,
    const long long* __restrict__ slot_mapping,
    int T, int Hkv, int page_size) {
  const long long total = (long long)T * Hkv * (PA_HEAD_DIM / 8);
  long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; idx < total; idx += stride) {
    const int s8 = (int)(idx % (PA_HEAD_DIM / 8));
    long long rest = idx / (PA_HEAD_DIM / 8);
    const int h = (int)(rest % Hkv);
    const int t = (int)(rest / Hkv);
    const long long slot = slot_mapping[t];
    if (slot < 0) continue;
    const long long page = slot / page_size;
    const long long off = slot % page_size;
    const long long src = (((long long)t * Hkv + h) * PA_HEAD_DIM) + s8 * 8;
    const long long dst =
        ((((long long)page * Hkv + h) * page_size + off) * PA_HEAD_DIM) + s8 * 8;
    *reinterpret_cast<uint4*>(k_cache + dst) =
        *reinterpret_cast<const uint4*>(k + src);
    *reinterpret_cast<uint4*>(v_cache + dst) =
        *reinterpret_cast<const uint4*>(v + src);
  }
}

// ---------------------------------------------------------------------------
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor block_tables, torch::Tensor seq_lens,
                            torch::Tensor ws_acc, torch::Tensor ws_ml,
                            double scale, long long chunk_keys) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && out.is_contiguous() &&
              k_cache.is_contiguous() && v_cache.is_contiguous());
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(seq_lens.scalar_type() == torch::kInt32);
  const int B = q.size(0);
  const int Hq = q.size(1);
  TORCH_CHECK(q.size(2) == PA_HEAD_DIM, "head dim must be 128");
  const int Hkv = k_cache.size(1);
  const int page_size = k_cache.size(2);
  const int G = Hq / Hkv;
  TORCH_CHECK(Hq % Hkv == 0 && G <= PA_MAX_GQA, "GQA group must be <= 16");
  TORCH_CHECK(chunk_keys % PA_KEYS_PER_PASS == 0);
  const int max_pages = block_tables.size(1);
  const int num_chunks = ws_ml.size(2);
  TORCH_CHECK(ws_acc.size(2) == num_chunks);

  hipStream_t stream = at::hip::getCurrentHIPStream();
  dim3 grid(B, Hkv, num_chunks);
  hipLaunchKernelGGL(paged_attention_chunk_kernel, grid, dim3(256), 0, stream,
                     ws_acc.data_ptr<float>(), ws_ml.data_ptr<float>(),
                     (const ushort*)q.data_ptr(),
                     (const ushort*)k_cache.data_ptr(),
                     (const ushort*)v_cache.data_ptr(),
                     block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(),
                     (float)scale, G, Hkv, page_size, max_pages,
                     (int)chunk_keys, num_chunks);
  hipLaunchKernelGGL(paged_attention_reduce_kernel, dim3(B, Hq), dim3(128), 0,
                     stream, (ushort*)out.data_ptr(), ws_acc.data_ptr<float>(),
                     ws_ml.data_ptr<float>(), seq_lens.data_ptr<int>(),
                     G, Hkv, (int)chunk_keys, num_chunks);
}

void reshape_and_cache(torch::Tensor k, torch::Tensor v,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor slot_mapping) {
  TORCH_CHECK(k.is_cuda() && k.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt64);
  TORCH_CHECK(k.is_contiguous() && v.is_contiguous());
  const int T = k.size(0);
  const int Hkv = k_cache.size(1);
  const int page_size = k_cache.size(2);
  long long total = (long long)T * Hkv * (PA_HEAD_DIM / 8);
  long long blocks = (total + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(reshape_and_cache_kernel, dim3((int)blocks), dim3(256), 0,
                     stream, (const ushort*)k.data_ptr(),
                     (const ushort*)v.data_ptr(), (ushort*)k_cache.data_ptr(),
                     (ushort*)v_cache.data_ptr(),
                     (const long long*)slot_mapping.data_ptr<int64_t>(),
                     T, Hkv, page_size);
}
