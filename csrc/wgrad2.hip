// Split-K transpose-A weight-grad GEMM, tiled like gemm2:
//   out(M, C) += A(K, M)^T @ B(K, C)    (fp32 out, bf16 in)
//
// Replaces csrc/wgrad.hip for 128-aligned shapes (transformer weight grads:
// M = out_features, C = in_features, K = tokens). The first version staged
// [k][m] tiles naturally and built transposed fragments with scalar LDS
// reads behind a per-32-K barrier — fully latency-serialized (105 us per
// 768x768xK=8192 call). Here both operands are register-staged TRANSPOSED
// into [m][k] / [c][k] XOR-swizzled images (glds cannot transpose), with
// the T14 split: next tile's global loads issue before the current tile's
// MFMAs, ds_writes land after the barrier. Fragments are then contiguous
// b128 reads and the inner loop is gemm2's.
//
// Geometry: M % 128 == 0, C % 128 == 0, K % 64 == 0.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdlib>

#define WAVE 64
#define WBM 128  // out rows (M) per block
#define WBC 128  // out cols (C) per block
#define WBK 64   // K rows per tile

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using uint4v = __attribute__((ext_vector_type(4))) unsigned int;

// each thread loads 4 x 16 B of one operand tile (64 K-rows x 128 cols):
// pass p: k = (t>>4) + 16p, col8 = (t&15)*8
__device__ __forceinline__ void w2_load(const bf16* __restrict__ src, long ld,
                                        int k0, int col0, int tid,
                                        uint4v regs[4], int k_lim) {
  const int kr = tid >> 4;
  const int c8 = (tid & 15) * 8;
  // a per-element bound check makes hipcc branch around each load and
  // drain vmcnt per element (guide §5 trap (c)); hoist to one uniform test
  if (k0 + 64 <= k_lim) {
#pragma unroll
    for (int p = 0; p < 4; ++p)
      regs[p] = *reinterpret_cast<const uint4v*>(
          src + (long)(k0 + kr + 16 * p) * ld + col0 + c8);
  } else {
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int k = k0 + kr + 16 * p;
      regs[p] = {};
      if (k < k_lim)
        regs[p] = *reinterpret_cast<const uint4v*>(src + (long)k * ld + col0 + c8);
    }
  }
}

// csum (optional): column sums of A (bias grads) accumulated by the
// blockIdx.y == 0 blocks from the staged subtiled image — lets the GGNN
// batched weight grads (K = steps*nodes, any K) use this kernel without a
// separate 60 MB colsum pass (same fold as csrc/wgrad.hip)
__global__ __launch_bounds__(256) void wgrad2_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    float* __restrict__ out, float* __restrict__ csum, int K, int M, int C,
    int kchunk, int zsplit) {
  // Natural [k][col] staging (coalesced 16-B LDS writes; the transposed
  // write variant serialized on 2 banks), transposed FRAGMENT READS as
  // scalar LDS loads (~4-way), double-buffered with the T14 load split so
  // the HBM latency hides under the MFMAs.
  extern __shared__ __attribute__((aligned(16))) char smem[];
// Subtiled image for ds_read_b64_tr_b16 hardware-transpose reads (guide
// T10; mapping verified by tools/tr_probe): [4 k][16 col] row-major
// subtiles, padded 128->144 B so the 16 staging b128 writes (subtile
// stride * 2 lanes/subtile) spread over the 64-dword bank modulus.
// One tr read delivers 4 k-column elements per lane (lane l&15 = column)
// — 2 reads per 8-deep MFMA fragment instead of 8 scalar ds_read_u16.
#define W2_SUBB 144
#define W2_SUB(kb, cb) (((kb)*8 + (cb)) * W2_SUBB)
#define W2_TILEB (16 * 8 * W2_SUBB)  // 64 k x 128 col = 18432 B
#define A_LDS(i) (smem + (i) * (2 * W2_TILEB))
#define B_LDS(i) (smem + W2_TILEB + (i) * (2 * W2_TILEB))

using bf16x4v = __attribute__((ext_vector_type(4))) __bf16;
typedef __attribute__((address_space(3))) bf16x4v lds_bf16x4;
  const int m0 = blockIdx.x * WBM;
  const int c0 = blockIdx.y * WBC;
  const int k_begin = blockIdx.z * kchunk;
  const int k_end = min(K, k_begin + kchunk);
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  const int wm = wid >> 1;
  const int wc = wid & 1;

  f32x4 acc[4][4] = {};
  float cs_acc = 0.f;
  const bool do_csum = (csum != nullptr) && (blockIdx.y == 0) && (tid < 128);

  // subtiled write: thread covers (k = tid>>4 + 16p, cols (tid&15)*8..+7)
  // -> one 16-B write into half ((tid&15)&1) of subtile (k>>2, (tid&15)>>1)
  auto write_nat = [&](char* lds, const uint4v regs[4]) {
    const int kr = tid >> 4;
    const int cb = (tid & 15) >> 1;
    const int halfb = ((tid & 15) & 1) * 16;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int k = kr + 16 * p;
      *reinterpret_cast<uint4v*>(lds + W2_SUB(k >> 2, cb) + (k & 3) * 32 +
                                 halfb) = regs[p];
    }
  };

  uint4v a_regs[4], b_regs[4];
  w2_load(A, M, k_begin, m0, tid, a_regs, k_end);
  w2_load(B, C, k_begin, c0, tid, b_regs, k_end);
  write_nat(A_LDS(0), a_regs);
  write_nat(B_LDS(0), b_regs);
  __syncthreads();

  int cur = 0;
  for (int k0 = k_begin; k0 < k_end; k0 += WBK) {
    const bool has_next = (k0 + WBK) < k_end;
    if (has_next) {  // T14: issue next tile's loads before this tile's MFMAs
      w2_load(A, M, k0 + WBK, m0, tid, a_regs, k_end);
      w2_load(B, C, k0 + WBK, c0, tid, b_regs, k_end);
    }
    if (do_csum) {
      const int klim = min(WBK, k_end - k0);
      const int cb = tid >> 4, cc = (tid & 15) * 2;
#pragma unroll 8
      for (int k = 0; k < klim; ++k)
        cs_acc += __bfloat162float(*reinterpret_cast<const __bf16*>(
            A_LDS(cur) + W2_SUB(k >> 2, cb) + (k & 3) * 32 + cc));
    }
    // ks stays a ROLLED loop: fully unrolling doubled the inner body and
    // the kernel ran at 34% SQ_WAIT_INST_ANY (instruction-fetch starved)
#pragma unroll 1
    for (int ks = 0; ks < 2; ++ks) {
      const int kb0 = ks * 8 + (lane >> 4) * 2;  // first 4-k subtile row
      const int slot = (lane & 15) * 8;          // this lane's 8-B column slot
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int cba = wm * 4 + f;
        const int cbb = wc * 4 + f;
        bf16x4v* af = reinterpret_cast<bf16x4v*>(&a_frag[f]);
        bf16x4v* bf = reinterpret_cast<bf16x4v*>(&b_frag[f]);
        af[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)(A_LDS(cur) + W2_SUB(kb0, cba) + slot));
        af[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)(A_LDS(cur) + W2_SUB(kb0 + 1, cba) + slot));
        bf[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)(B_LDS(cur) + W2_SUB(kb0, cbb) + slot));
        bf[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)(B_LDS(cur) + W2_SUB(kb0 + 1, cbb) + slot));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fc = 0; fc < 4; ++fc)
          acc[fm][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fc], acc[fm][fc], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
    if (has_next) {
      write_nat(A_LDS(cur ^ 1), a_regs);
      write_nat(B_LDS(cur ^ 1), b_regs);
    }
    __syncthreads();
    cur ^= 1;
  }

#undef A_LDS
#undef B_LDS
  if (do_csum) {
    const int m = m0 + (tid >> 4) * 16 + (tid & 15);
    if (m < M) atomicAdd(csum + m, cs_acc);
  }
  // epilogue: D col = lane&15 (C dim), row = (lane>>4)*4 + i (M dim)
#pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
    const int m_base = m0 + wm * 64 + fm * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int fc = 0; fc < 4; ++fc) {
      const int c = c0 + wc * 64 + fc * 16 + (lane & 15);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        if (zsplit == 1)
          out[(long)(m_base + i) * C + c] = acc[fm][fc][i];
        else
          atomicAdd(out + (long)(m_base + i) * C + c, acc[fm][fc][i]);
      }
    }
  }
}

void launch_wgrad2(const bf16* A, const bf16* B, float* out, float* csum,
                   int K, int M, int C, int accumulate, hipStream_t stream) {
  const int tiles = (M / WBM) * (C / WBC);
  // K-chunk floor trades chip fill (more z-chunks) against fp32 atomic
  // output traffic (each chunk adds one full M x C atomic pass); 768 won
  // the sweep at the square 768 shapes (27.2 -> 23.7 us at K=4096) and at
  // K=8192, fat shapes indifferent (tools/wg2_probe.py)
  int kfloor = 768;
  if (const char* e = getenv("DFA_WG2_KFLOOR")) kfloor = atoi(e);
  int zsplit = max(1, 512 / tiles);
  int kchunk = (K + zsplit - 1) / zsplit;
  kchunk = ((kchunk + WBK - 1) / WBK) * WBK;
  if (kchunk < kfloor) kchunk = min(((K + WBK - 1) / WBK) * WBK, kfloor);
  zsplit = (K + kchunk - 1) / kchunk;
  const dim3 grid(M / WBM, C / WBC, zsplit);
  // accumulate: out is a live accumulator (the flat .grad view) — force the
  // atomic epilogue even when a single K chunk would have plain-stored
  const int zarg = accumulate ? max(zsplit, 2) : zsplit;
  hipLaunchKernelGGL(wgrad2_kernel, grid, dim3(256), 73728, stream, A, B, out,
                     csum, K, M, C, kchunk, zarg);
}
