// Flash-style causal prefill attention for gfx950 (MI355X), varlen batch,
// GQA-aware.
//
// Replaces the library-GEMM composition (which materializes S x S score
// tiles and loops sequences on the host).  Structure per the CDNA4 guide
// §Appendix B "Fused attention prefill":
//
//   grid = (q_tiles_total, Hq) x 256 threads (4 waves)
//   Each workgroup owns one 64-row Q tile of one sequence for one q head
//   (wave w handles rows w*16..w*16+15) and iterates 32-key K/V tiles:
//     QK^T via mfma_f32_16x16x32_bf16 over D=128 (4 MFMA per 16-key tile)
//     online softmax per q row (running m, l) with causal masking
//     P staged through LDS (bf16) to re-fragment for the PV MFMA
//     PV accumulates O[16 x 128] in registers, rescaled per tile
//   Q staged in LDS once; K/V tiles staged cooperatively (coalesced); all
//   LDS tiles use padded pitches to keep ds_read conflicts <= 2-way (G4).
//
// Fragment layouts as verified for skinny_gemm (tests + asymmetric rule):
//   A frag: lane l holds A[m = l%16][k = (l/16)*8 + j]
//   B frag: lane l holds B[k = (l/16)*8 + j][n = l%16]
//   C/D:    lane l reg r holds D[row = (l/16)*4 + r][col = l%16]
#include "common.h"

using bf16x8_t = __attribute__((ext_vector_type(8))) short;
using f32x4_t = __attribute__((ext_vector_type(4))) float;

#define PF_QT 64          // q rows per workgroup
#define PF_KT 32          // keys per tile
#define PF_D 128
#define PF_DPITCH (PF_D + 8)
#define PF_PPITCH (PF_KT + 8)

__global__ __launch_bounds__(256)
void prefill_attention_kernel(
    ushort* __restrict__ out,          // [T, Hq*128] contiguous
    const ushort* __restrict__ q,      // [T, Hq, 128], row stride q_stride
    const ushort* __restrict__ k,      // [T, Hkv, 128], row stride k_stride
    const ushort* __restrict__ v,      // [T, Hkv, 128], row stride v_stride
    const int* __restrict__ tile_seq,  // [tiles] sequence index
    const int* __restrict__ tile_q0,   // [tiles] first q row (seq-local)
    const int* __restrict__ seq_starts,  // [B+1] global q-token offsets
    const int* __restrict__ kv_starts,   // [B+1] global k/v row offsets
    const int* __restrict__ q_offs,      // [B] cached keys before the chunk
    float scale, int Hq, int Hkv,
    long long q_stride, long long k_stride, long long v_stride) {
  const int tile = blockIdx.x;
  const int hq = blockIdx.y;
  const int hkv = hq / (Hq / Hkv);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int frag_row = lane % 16;
  const int frag_kgrp = lane / 16;

  const int seq = tile_seq[tile];
  const int s0 = seq_starts[seq];
  const int S = seq_starts[seq + 1] - s0;
  const int q0 = tile_q0[tile];              // seq-local first q row of tile
  const int q_rows = min(PF_QT, S - q0);
  const int q_end = q0 + q_rows - 1;         // last (seq-local) q position
  // chunked prefill: the chunk's q rows sit AFTER `off` cached keys; the
  // k/v rows for this sequence span the whole context (prefix + chunk)
  const int kv0 = kv_starts[seq];
  const int KV = kv_starts[seq + 1] - kv0;
  const int off = q_offs[seq];

  __shared__ ushort q_lds[PF_QT][PF_DPITCH];
  __shared__ ushort k_lds[PF_KT][PF_DPITCH];
  __shared__ ushort v_lds[PF_KT][PF_DPITCH];
  __shared__ ushort p_lds[4][16][PF_PPITCH];

  // ---- stage Q tile (scaled path keeps bf16; scale applied to S) ----
  for (int u = threadIdx.x; u < PF_QT * (PF_D / 8); u += blockDim.x) {
    const int row = u / (PF_D / 8);
    const int col = (u % (PF_D / 8)) * 8;
    uint4 val = make_uint4(0, 0, 0, 0);
    if (row < q_rows)
      val = *reinterpret_cast<const uint4*>(
          q + (long long)(s0 + q0 + row) * q_stride
          + (long long)hq * PF_D + col);
    *reinterpret_cast<uint4*>(&q_lds[row][col]) = val;
  }
  __syncthreads();

  // per-wave state: rows wave*16 .. wave*16+15; this lane owns 4 of them
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -INFINITY;
    l_run[r] = 0.0f;
  }
  f32x4_t o_acc[PF_D / 16];                 // O[4 rows][8 d-tiles x 16]
#pragma unroll
  for (int n = 0; n < PF_D / 16; ++n) o_acc[n] = {0.f, 0.f, 0.f, 0.f};

  // Q fragments for this wave, loaded once (4 d-chunks of 32)
  bf16x8_t qfrag[4];
#pragma unroll
  for (int dc = 0; dc < 4; ++dc) {
    bf16x8 tmp;
    tmp.u = *reinterpret_cast<const uint4*>(
        &q_lds[wave * 16 + frag_row][dc * 32 + frag_kgrp * 8]);
#pragma unroll
    for (int j = 0; j < 8; ++j) qfrag[dc][j] = (short)tmp.h[j];
  }

  for (int k0 = 0; k0 <= off + q_end; k0 += PF_KT) {
    const int kw = min(PF_KT, KV - k0);
    // ---- stage K and V tiles (coalesced; invalid rows zero) ----
    __syncthreads();
    for (int u = threadIdx.x; u < PF_KT * (PF_D / 8); u += blockDim.x) {
      const int row = u / (PF_D / 8);
      const int col = (u % (PF_D / 8)) * 8;
      uint4 kv_ = make_uint4(0, 0, 0, 0);
      uint4 vv_ = make_uint4(0, 0, 0, 0);
      if (row < kw) {
        const long long tok = (long long)(kv0 + k0 + row);
        kv_ = *reinterpret_cast<const uint4*>(
            k + tok * k_stride + (long long)hkv * PF_D + col);
        vv_ = *reinterpret_cast<const uint4*>(
            v + tok * v_stride + (long long)hkv * PF_D + col);
      }
      *reinterpret_cast<uint4*>(&k_lds[row][col]) = kv_;
      *reinterpret_cast<uint4*>(&v_lds[row][col]) = vv_;
    }
    __syncthreads();

    // ---- S = Q K^T for this wave's 16 rows x PF_KT keys ----
    f32x4_t s_acc[PF_KT / 16];
#pragma unroll
    for (int nt = 0; nt < PF_KT / 16; ++nt) {
      s_acc[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int dc = 0; dc < 4; ++dc) {
        // B fragment: K^T[d][key] = K[key][d]
        bf16x8_t bfrag;
        bf16x8 tmp;
        tmp.u = *reinterpret_cast<const uint4*>(
            &k_lds[nt * 16 + frag_row][dc * 32 + frag_kgrp * 8]);
#pragma unroll
        for (int j = 0; j < 8; ++j) bfrag[j] = (short)tmp.h[j];
        // NOTE: for QK^T we need contraction over d with A=Q rows.  The
        // B-frag layout wants B[d][key]: lane l holds K[key=l%16][d-chunk]
        // which IS B^T — mfma(A,B) computes A·B with B[k][n]; feeding
        // K-rows as B fragments yields S[q][key] = sum_d Q[q][d]K[key][d].
        s_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qfrag[dc], bfrag, s_acc[nt], 0, 0, 0);
      }
    }

    // ---- causal mask + online softmax (rows (l/16)*4+r, col l%16) ----
    float p_val[PF_KT / 16][4];
    float m_new[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) m_new[r] = m_run[r];
#pragma unroll
    for (int nt = 0; nt < PF_KT / 16; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        // D-side row within the tile: this wave's block + lane mapping
        const int row = wave * 16 + frag_kgrp * 4 + r;
        const int key = k0 + nt * 16 + frag_row;  // this lane's column
        float sv = s_acc[nt][r] * scale;
        if (key > off + q0 + row || key >= KV || row >= q_rows)
          sv = -INFINITY;
        p_val[nt][r] = sv;
      }
    }
    // row-max across the 16 columns (lanes sharing frag_kgrp)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = -INFINITY;
#pragma unroll
      for (int nt = 0; nt < PF_KT / 16; ++nt) mx = fmaxf(mx, p_val[nt][r]);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, WAVE_SIZE));
      m_new[r] = fmaxf(m_run[r], mx);
    }
    // P = exp(S - m_new); row-sum; rescale running state
    float rescale[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      rescale[r] = (m_run[r] == -INFINITY || m_new[r] == -INFINITY)
          ? 0.0f : __expf(m_run[r] - m_new[r]);
      float lsum = 0.0f;
#pragma unroll
      for (int nt = 0; nt < PF_KT / 16; ++nt) {
        float e = (p_val[nt][r] == -INFINITY || m_new[r] == -INFINITY)
            ? 0.0f : __expf(p_val[nt][r] - m_new[r]);
        p_val[nt][r] = e;
        lsum += e;
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        lsum += __shfl_xor(lsum, off, WAVE_SIZE);
      l_run[r] = l_run[r] * rescale[r] + lsum;
      m_run[r] = m_new[r];
    }

    // ---- stage P (bf16) for the PV re-fragmentation ----
#pragma unroll
    for (int nt = 0; nt < PF_KT / 16; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p_lds[wave][frag_kgrp * 4 + r][nt * 16 + frag_row] =
            f32_to_bf16(p_val[nt][r]);
      }
    }
    // O rescale while P lands in LDS
#pragma unroll
    for (int n = 0; n < PF_D / 16; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[n][r] *= rescale[r];
    }
    __syncthreads();  // P visible; V staged earlier this iteration

    // ---- O += P V : A = P[16 q x 32 k], B = V[32 k x 16 d] ----
    bf16x8_t pfrag;
    {
      bf16x8 tmp;
      tmp.u = *reinterpret_cast<const uint4*>(
          &p_lds[wave][frag_row][frag_kgrp * 8]);
#pragma unroll
      for (int j = 0; j < 8; ++j) pfrag[j] = (short)tmp.h[j];
    }
#pragma unroll
    for (int n = 0; n < PF_D / 16; ++n) {
      bf16x8_t vfrag;
      bf16x8 tmp;
      // B[k][n]: lane l holds V[k = frag_kgrp*8+j][d = n*16 + frag_row]
      // -> strided LDS reads (d fixed per lane, k varies with j)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        tmp.h[j] = v_lds[frag_kgrp * 8 + j][n * 16 + frag_row];
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) vfrag[j] = (short)tmp.h[j];
      o_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag,
                                                         o_acc[n], 0, 0, 0);
    }
  }

  // ---- write O / l  (rows (l/16)*4+r of this wave's 16-row block) ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = wave * 16 + frag_kgrp * 4 + r;
    if (row >= q_rows) continue;
    const float inv = 1.0f / fmaxf(l_run[r], 1e-20f);
    ushort* orow = out + (long long)(s0 + q0 + row) * (Hq * PF_D)
        + (long long)hq * PF_D;
#pragma unroll
    for (int n = 0; n < PF_D / 16; ++n) {
      orow[n * 16 + frag_row] = f32_to_bf16(o_acc[n][r] * inv);
    }
  }
}

// ---------------------------------------------------------------------------
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void prefill_attention(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                       torch::Tensor v, torch::Tensor tile_seq,
                       torch::Tensor tile_q0, torch::Tensor seq_starts,
                       torch::Tensor kv_starts, torch::Tensor q_offs,
                       double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == PF_D);
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == PF_D);
  TORCH_CHECK(v.stride(2) == 1 && v.stride(1) == PF_D);
  TORCH_CHECK(tile_seq.scalar_type() == torch::kInt32 &&
              tile_q0.scalar_type() == torch::kInt32 &&
              seq_starts.scalar_type() == torch::kInt32 &&
              kv_starts.scalar_type() == torch::kInt32 &&
              q_offs.scalar_type() == torch::kInt32);
  const int Hq = q.size(1);
  const int Hkv = k.size(1);
  TORCH_CHECK(q.size(2) == PF_D, "head dim must be 128");
  TORCH_CHECK(Hq % Hkv == 0);
  const int tiles = tile_seq.size(0);
  if (tiles == 0) return;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(prefill_attention_kernel, dim3(tiles, Hq), dim3(256), 0,
                     stream, (ushort*)out.data_ptr(),
                     (const ushort*)q.data_ptr(), (const ushort*)k.data_ptr(),
                     (const ushort*)v.data_ptr(), tile_seq.data_ptr<int>(),
                     tile_q0.data_ptr<int>(), seq_starts.data_ptr<int>(),
                     kv_starts.data_ptr<int>(), q_offs.data_ptr<int>(),
                     (float)scale, Hq, Hkv, (long long)q.stride(0),
                     (long long)k.stride(0), (long long)v.stride(0));
}
