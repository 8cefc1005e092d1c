// Split-K weight-gradient GEMM: out(M, C) += A(K, M)^T @ [B1|B2](K, C).
//
// The transpose-A reduction GEMMs (weight gradients, K = nodes*steps ~ 58k,
// output 512x256 or 128x128) are where hipBLASLt collapses on these shapes:
// its heuristic picks a kernel with (M/MT)*(C/NT) = 6 workgroups and no
// K-split — 2.3% of the chip, 87 us (tools/gemm_probe.py). This kernel
// splits K across blockIdx.z with fp32 atomicAdd epilogues.
//
// Geometry: 128x128 output tile per block, 4 waves (2x2) of 64x64 each,
// v_mfma_f32_16x16x32_bf16. A-tiles are staged [k][m] in LDS exactly as
// they lie in memory (coalesced 16B) and the transposed fragments are built
// with scalar LDS reads — at 16 MFMAs per 8-fragment k-step the MFMA issue
// rate covers the scalar-read issue cost (the 64x64-wave shape is what
// makes that ratio work; at 32x32 the kernel was LDS-issue bound).
//
// B is split (B1 cols [0,C1), B2 cols [C1,C)) so gWcat = grad_gicat^T @
// [messages | hidden] needs no concatenation.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define KS 32  // K per MFMA step

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using uint4v = __attribute__((ext_vector_type(4))) unsigned int;

// transposed-fragment bank map: at a 256-B row stride all four (lane>>4)
// k-groups (8 rows = 512 dwords = 0 mod 32) hit the same 8 banks; XOR the
// column byte with ((k>>3)&3)<<5 (bits 5-6 only: bijective inside the row,
// keeps 16-B write alignment, disjoint bank blocks per k-group) — the same
// fix as csrc/wgrad2.hip, worth 4.07e6 LDS_BANK_CONFLICT/call here
#define W1_ADDR(k, colbyte) ((k)*256 + ((colbyte) ^ ((((k) >> 3) & 3) << 5)))

// csum (optional): column sums of A (the bias gradients) accumulated by
// the blockIdx.y == 0 blocks from the already-staged LDS tiles — removes
// the separate colsum pass over the same 60 MB of activation grads
// GRU scatter epilogue (optional): when out_ih != nullptr the (4H, 2H)
// gWcat tile output is scattered straight into the torch-GRUCell grad
// layouts instead of one packed buffer — gW_ih (3H, H) = rows [0,3H) x
// m-cols, gW_hh (3H, H) = rows [0,2H) + [3H,4H) x h-cols (the [2H,3H)
// h-block is structurally zero), and the bias column-sums go to b_ih
// (rows [0,3H)) and b_hh (rows [0,2H) + [3H,4H)), rows [0,2H) feeding
// BOTH biases (r/z gates add b_ih and b_hh). Replaces the narrow/cat/
// AccumulateGrad chain in the flat-optimizer path (ops/flowgnn.py).
__global__ __launch_bounds__(256) void wgrad_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B1,
    const bf16* __restrict__ B2, float* __restrict__ out,
    float* __restrict__ csum, int K, int M, int C, int C1, int kchunk,
    float* __restrict__ out_ih, float* __restrict__ out_hh,
    float* __restrict__ cs_ih, float* __restrict__ cs_hh) {
  __shared__ char lds_a[KS * 256];
  __shared__ char lds_b[KS * 256];

  const int m0 = blockIdx.x * 128;
  const int c0 = blockIdx.y * 128;
  const int k_begin = blockIdx.z * kchunk;
  const int k_end = min(K, k_begin + kchunk);
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  const int wm = wid >> 1;  // 2 row-waves (m), 2 col-waves (c)
  const int wc = wid & 1;

  f32x4 acc[4][4] = {};
  float cs_acc = 0.f;
  const bool do_csum = (csum != nullptr || cs_ih != nullptr) &&
                       (blockIdx.y == 0) && (tid < 128);

  for (int k0 = k_begin; k0 < k_end; k0 += KS) {
    // stage [KS][128] tiles (row-major, coalesced 16B; 16 threads/row)
    {
      const int row = tid >> 4;          // 0..15
      const int coff = (tid & 15) * 8;   // element offset in tile row
#pragma unroll
      for (int rr = 0; rr < KS; rr += 16) {
        const int k = k0 + row + rr;
        uint4v va = {}, vb = {};
        if (k < k_end) {
          const int mg = m0 + coff;
          if (mg < M) va = *reinterpret_cast<const uint4v*>(A + (long)k * M + mg);
          const int cg = c0 + coff;
          if (cg < C) {
            const bf16* src = (cg < C1) ? (B1 + (long)k * C1 + cg)
                                        : (B2 + (long)k * (C - C1) + (cg - C1));
            vb = *reinterpret_cast<const uint4v*>(src);
          }
        }
        *reinterpret_cast<uint4v*>(lds_a + W1_ADDR(row + rr, coff * 2)) = va;
        *reinterpret_cast<uint4v*>(lds_b + W1_ADDR(row + rr, coff * 2)) = vb;
      }
    }
    __syncthreads();

    if (do_csum) {
      const int klim = min(KS, k_end - k0);
#pragma unroll 8
      for (int k = 0; k < klim; ++k)
        cs_acc += __bfloat162float(
            *reinterpret_cast<const __bf16*>(lds_a + W1_ADDR(k, tid * 2)));
    }

    bf16x8 a_frag[4], b_frag[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      const int m_loc = wm * 64 + f * 16 + (lane & 15);
      const int c_loc = wc * 64 + f * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int k = (lane >> 4) * 8 + j;
        a_frag[f][j] = *reinterpret_cast<const __bf16*>(lds_a + W1_ADDR(k, m_loc * 2));
        b_frag[f][j] = *reinterpret_cast<const __bf16*>(lds_b + W1_ADDR(k, c_loc * 2));
      }
    }
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fc = 0; fc < 4; ++fc)
        acc[fm][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[fm], b_frag[fc],
                                                              acc[fm][fc], 0, 0, 0);
    __syncthreads();
  }

  if (do_csum && m0 + tid < M) {
    const int m = m0 + tid;
    if (cs_ih != nullptr) {
      const int Hg = C1;  // gate width: C1 == H for the GGNN gWcat call
      if (m < 3 * Hg) atomicAdd(cs_ih + m, cs_acc);
      if (m < 2 * Hg) atomicAdd(cs_hh + m, cs_acc);
      else if (m >= 3 * Hg) atomicAdd(cs_hh + m - Hg, cs_acc);
    } else {
      atomicAdd(csum + m, cs_acc);
    }
  }
  // epilogue: D col = lane&15, row = (lane>>4)*4 + i; atomic fp32 accumulate
#pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
    const int m_base = m0 + wm * 64 + fm * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int fc = 0; fc < 4; ++fc) {
      const int c = c0 + wc * 64 + fc * 16 + (lane & 15);
      if (c >= C) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int m = m_base + i;
        if (m >= M) continue;
        if (out_ih != nullptr) {
          const int Hg = C1;
          if (c < Hg) {
            if (m < 3 * Hg) atomicAdd(out_ih + (long)m * Hg + c, acc[fm][fc][i]);
          } else {
            const int ch = c - Hg;
            if (m < 2 * Hg) atomicAdd(out_hh + (long)m * Hg + ch, acc[fm][fc][i]);
            else if (m >= 3 * Hg)
              atomicAdd(out_hh + (long)(m - Hg) * Hg + ch, acc[fm][fc][i]);
          }
        } else {
          atomicAdd(out + (long)m * C + c, acc[fm][fc][i]);
        }
      }
    }
  }
}

void launch_wgrad(const bf16* A, const bf16* B1, const bf16* B2, float* out,
                  float* csum, int K, int M, int C, int C1, hipStream_t stream,
                  float* out_ih, float* out_hh, float* cs_ih, float* cs_hh) {
  // size the K-split so the grid comfortably fills 256 CUs
  const int tiles = ((M + 127) / 128) * ((C + 127) / 128);
  int zsplit = max(1, 512 / tiles);
  int kchunk = (K + zsplit - 1) / zsplit;
  kchunk = ((kchunk + KS - 1) / KS) * KS;
  // keep >= 1 KiB-ish of K per block: beyond that the fp32 atomic epilogues
  // (zsplit adds per output word) outweigh the extra parallelism
  if (kchunk < 1024) kchunk = min(((K + KS - 1) / KS) * KS, 1024);
  zsplit = (K + kchunk - 1) / kchunk;
  const dim3 grid((M + 127) / 128, (C + 127) / 128, zsplit);
  hipLaunchKernelGGL(wgrad_kernel, grid, dim3(256), 0, stream, A, B1, B2, out,
                     csum, K, M, C, C1, kchunk, out_ih, out_hh, cs_ih, cs_hh);
}
