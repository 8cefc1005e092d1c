// MFMA bf16 GEMM with bias + split-A, tuned for the flow-GNN's node-state
// GEMMs on MI355X (gfx950): out(N, COL) = [A1|A2](N, K) @ W(COL, K)^T + bias.
//
// Shapes in this framework: N ~ 11-12k nodes, K in {128, 256}, COL in
// {128, 512}. hipBLASLt's heuristic picks an 84-91us kernel for these
// shapes (rocprof, profiles/); this kernel is ~10x faster by being sized
// for them: 64x128 tile, BK=64, v_mfma_f32_16x16x32_bf16, XOR-swizzled LDS
// (cdna_hip_programming.md T2: byte ^= (row&7)<<4 spreads the 16-lane
// ds_read_b128 fragment groups across bank slots).
//
// The split-A form reads K-columns [0,K1) from A1 and [K1,K) from A2 so the
// fused-GRU GEMM can consume [messages | hidden] without materializing the
// concatenation.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;  // 4 VGPRs
using f32x4 = __attribute__((ext_vector_type(4))) float;
using uint4v = __attribute__((ext_vector_type(4))) unsigned int;

// tile geometry
#define BM 64
#define BN 128
#define BK 64
#define ROWB (BK * 2)  // bytes per LDS row (64 bf16)

__device__ __forceinline__ int swz(int row, int byte) {
  return row * ROWB + (byte ^ ((row & 7) << 4));
}

// out(N, COL) = concat_K(A1, A2)(N, K) @ W(COL, K)^T + bias
// grid = (ceil(N/BMT), COL/BN), block = 256 (4 waves, 2x2)
// BMT = 64 normally; 32 when the 64-row grid would underfill 256 CUs —
// these GNN GEMMs are HBM-latency-bound at ~0.7 waves/SIMD (PMC: WAIT_ANY
// 73% of wave cycles), so doubling the block count is the lever
template <int BMT>
__global__ __launch_bounds__(256) void gemm_bias_kernel(
    const bf16* __restrict__ A1, const bf16* __restrict__ A2,
    const bf16* __restrict__ W, const bf16* __restrict__ bias,
    const bf16* __restrict__ addend, long astride,
    const bf16* __restrict__ addend2, long astride2,
    bf16* __restrict__ out, int N, int K,
    int K1, int COL) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* a_lds = smem;                 // BMT x ROWB
  char* b_lds = smem + BMT * ROWB;    // BN x ROWB = 16 KiB

  const int r0 = blockIdx.x * BMT;
  const int c0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;           // 4 waves: (wm, wn) = (wid>>1, wid&1)
  const int wm = wid >> 1;            // 2 row-waves of 32 rows
  const int wn = wid & 1;             // 2 col-waves of 64 cols

  constexpr int MFRAG = BMT / 32;     // fragments per row-wave
  f32x4 acc[MFRAG][4] = {};

  // T14 double-buffered staging: next K-tile's global loads issue before
  // this tile's MFMAs (these N x 128-ish GEMMs are HBM-latency-bound at
  // ~1.4 blocks/CU; single-buffered staging serialized load latency with
  // the MFMA phase)
  const int srow = tid >> 3;         // 0..31
  const int soff = (tid & 7) * 16;   // byte offset in row
  constexpr int NAR = BMT / 32;      // A rows per thread
  auto load_regs = [&](int kk, uint4v ar[NAR], uint4v br[4]) {
#pragma unroll
    for (int p = 0; p < NAR; ++p) {
      const int rr = srow + p * 32;
      const int gr = r0 + rr;
      ar[p] = {};
      if (gr < N) {
        const int gk = kk + soff / 2;  // element index in K
        const bf16* src = (gk < K1) ? (A1 + (long)gr * K1 + gk)
                                    : (A2 + (long)gr * (K - K1) + (gk - K1));
        ar[p] = *reinterpret_cast<const uint4v*>(src);
      }
    }
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rr = srow + p * 32;
      br[p] = *reinterpret_cast<const uint4v*>(W + (long)(c0 + rr) * K + kk +
                                               soff / 2);
    }
  };
  auto write_regs = [&](char* ab, char* bb, const uint4v ar[NAR],
                        const uint4v br[4]) {
#pragma unroll
    for (int p = 0; p < NAR; ++p)
      *reinterpret_cast<uint4v*>(ab + swz(srow + p * 32, soff)) = ar[p];
#pragma unroll
    for (int p = 0; p < 4; ++p)
      *reinterpret_cast<uint4v*>(bb + swz(srow + p * 32, soff)) = br[p];
  };
#define GB_ABUF(i) (a_lds + (i) * (BMT + BN) * ROWB)
#define GB_BBUF(i) (b_lds + (i) * (BMT + BN) * ROWB)
  uint4v areg[NAR], breg[4];
  load_regs(0, areg, breg);
  write_regs(GB_ABUF(0), GB_BBUF(0), areg, breg);
  __syncthreads();
  int cur = 0;
  for (int kk = 0; kk < K; kk += BK) {
    const bool has_next = (kk + BK) < K;
    if (has_next) load_regs(kk + BK, areg, breg);
    char* a_lds_c = GB_ABUF(cur);
    char* b_lds_c = GB_BBUF(cur);

#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      // fragment k-range: lane kb = lane>>4 (4 groups of 8 elems)
      const int kbyte = ks * 64 + (lane >> 4) * 16;
      bf16x8 a_frag[MFRAG], b_frag[4];
#pragma unroll
      for (int m = 0; m < MFRAG; ++m) {
        const int row = wm * (BMT / 2) + m * 16 + (lane & 15);
        a_frag[m] = *reinterpret_cast<const bf16x8*>(a_lds_c + swz(row, kbyte));
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int row = wn * 64 + n * 16 + (lane & 15);
        b_frag[n] = *reinterpret_cast<const bf16x8*>(b_lds_c + swz(row, kbyte));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m = 0; m < MFRAG; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    if (has_next) write_regs(GB_ABUF(cur ^ 1), GB_BBUF(cur ^ 1), areg, breg);
    __syncthreads();
    cur ^= 1;
  }
#undef GB_ABUF
#undef GB_BBUF

  // epilogue: D mapping col = lane&15, row = (lane>>4)*4 + i
#pragma unroll
  for (int m = 0; m < MFRAG; ++m) {
    const int row_base = r0 + wm * (BMT / 2) + m * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int col = c0 + wn * 64 + n * 16 + (lane & 15);
      const float b = bias ? __bfloat162float(bias[col]) : 0.0f;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = row_base + i;
        if (row < N) {
          float v = acc[m][n][i] + b;
          if (addend) v += __bfloat162float(addend[(long)row * astride + col]);
          if (addend2) v += __bfloat162float(addend2[(long)row * astride2 + col]);
          out[(long)row * COL + col] = __float2bfloat16(v);
        }
      }
    }
  }
}

void launch_gemm_bias2(const bf16* A1, const bf16* A2, const bf16* W,
                       const bf16* bias, const bf16* addend, long astride,
                       const bf16* addend2, long astride2, bf16* out, int N,
                       int K, int K1, int COL, hipStream_t stream) {
  const long blocks64 = (long)((N + BM - 1) / BM) * (COL / BN);
  if (blocks64 < 384) {
    const dim3 grid((N + 31) / 32, COL / BN);
    const size_t lds = 2 * (32 + BN) * ROWB;
    hipLaunchKernelGGL(gemm_bias_kernel<32>, grid, dim3(256), lds, stream, A1,
                       A2, W, bias, addend, astride, addend2, astride2, out, N,
                       K, K1, COL);
  } else {
    const dim3 grid((N + BM - 1) / BM, COL / BN);
    const size_t lds = 2 * (BM + BN) * ROWB;
    hipLaunchKernelGGL(gemm_bias_kernel<BM>, grid, dim3(256), lds, stream, A1,
                       A2, W, bias, addend, astride, addend2, astride2, out, N,
                       K, K1, COL);
  }
}

void launch_gemm_bias(const bf16* A1, const bf16* A2, const bf16* W,
                      const bf16* bias, const bf16* addend, bf16* out, int N,
                      int K, int K1, int COL, hipStream_t stream) {
  launch_gemm_bias2(A1, A2, W, bias, addend, COL, nullptr, 0, out, N, K, K1,
                    COL, stream);
}

// ---------------------------------------------------------------------------
// Fused gate GEMM + GRU cell: out of the Wcat GEMM's accumulators, compute
// r/z/n and the blended h' directly (VERDICT round-1 item 4: "fuse
// gru_gates into the Wcat GEMM epilogue"). Uses a gate-INTERLEAVED weight
// layout Wcat_perm[row j*4+g] = Wcat[row g*H+j] so one 128-col block owns
// all four gate pre-activations for 32 j-columns; the accumulator tile is
// bounced through LDS in fp32 (more precise than the old bf16 gicat
// round-trip) and the gate math runs as the epilogue. Emits h_new plus the
// R/Z/Nn/HN activations the backward consumes (gru_gates2_bwd unchanged).
// ---------------------------------------------------------------------------

__device__ __forceinline__ float sigf_(float x) { return 1.f / (1.f + __expf(-x)); }

template <int BMT>
__global__ __launch_bounds__(256) void gemm_gru_kernel(
    const bf16* __restrict__ A1, const bf16* __restrict__ A2,
    const bf16* __restrict__ Wperm, const bf16* __restrict__ bperm,
    const bf16* __restrict__ h_in, bf16* __restrict__ h_new,
    bf16* __restrict__ R, bf16* __restrict__ Z, bf16* __restrict__ Nn,
    bf16* __restrict__ HN, int N, int K, int K1, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* a_lds = smem;
  char* b_lds = smem + BMT * ROWB;

  const int r0 = blockIdx.x * BMT;
  const int c0 = blockIdx.y * BN;  // column block of the PERMUTED 4H space
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  const int wm = wid >> 1;
  const int wn = wid & 1;

  constexpr int MFRAG = BMT / 32;
  f32x4 acc[MFRAG][4] = {};

  // same T14 register double-buffering as gemm_bias_kernel above
  const int srow = tid >> 3;
  const int soff = (tid & 7) * 16;
  constexpr int NAR = BMT / 32;
  auto load_regs = [&](int kk, uint4v ar[NAR], uint4v br[4]) {
#pragma unroll
    for (int p = 0; p < NAR; ++p) {
      const int rr = srow + p * 32;
      const int gr = r0 + rr;
      ar[p] = {};
      if (gr < N) {
        const int gk = kk + soff / 2;
        const bf16* src = (gk < K1) ? (A1 + (long)gr * K1 + gk)
                                    : (A2 + (long)gr * (K - K1) + (gk - K1));
        ar[p] = *reinterpret_cast<const uint4v*>(src);
      }
    }
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rr = srow + p * 32;
      br[p] = *reinterpret_cast<const uint4v*>(Wperm + (long)(c0 + rr) * K +
                                               kk + soff / 2);
    }
  };
  auto write_regs = [&](char* ab, char* bb, const uint4v ar[NAR],
                        const uint4v br[4]) {
#pragma unroll
    for (int p = 0; p < NAR; ++p)
      *reinterpret_cast<uint4v*>(ab + swz(srow + p * 32, soff)) = ar[p];
#pragma unroll
    for (int p = 0; p < 4; ++p)
      *reinterpret_cast<uint4v*>(bb + swz(srow + p * 32, soff)) = br[p];
  };
#define GB_ABUF(i) (a_lds + (i) * (BMT + BN) * ROWB)
#define GB_BBUF(i) (b_lds + (i) * (BMT + BN) * ROWB)
  uint4v areg[NAR], breg[4];
  load_regs(0, areg, breg);
  write_regs(GB_ABUF(0), GB_BBUF(0), areg, breg);
  __syncthreads();
  int cur = 0;
  for (int kk = 0; kk < K; kk += BK) {
    const bool has_next = (kk + BK) < K;
    if (has_next) load_regs(kk + BK, areg, breg);
    char* a_lds_c = GB_ABUF(cur);
    char* b_lds_c = GB_BBUF(cur);
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      const int kbyte = ks * 64 + (lane >> 4) * 16;
      bf16x8 a_frag[MFRAG], b_frag[4];
#pragma unroll
      for (int m = 0; m < MFRAG; ++m) {
        const int row = wm * (BMT / 2) + m * 16 + (lane & 15);
        a_frag[m] = *reinterpret_cast<const bf16x8*>(a_lds_c + swz(row, kbyte));
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int row = wn * 64 + n * 16 + (lane & 15);
        b_frag[n] = *reinterpret_cast<const bf16x8*>(b_lds_c + swz(row, kbyte));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m = 0; m < MFRAG; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    if (has_next) write_regs(GB_ABUF(cur ^ 1), GB_BBUF(cur ^ 1), areg, breg);
    __syncthreads();
    cur ^= 1;
  }
#undef GB_ABUF
#undef GB_BBUF

  // bounce fp32 gate pre-activations (+bias) through LDS, row stride 132
  float* tile = reinterpret_cast<float*>(smem);
#pragma unroll
  for (int m = 0; m < MFRAG; ++m) {
    const int row = wm * (BMT / 2) + m * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int col = wn * 64 + n * 16 + (lane & 15);
      const float b = __bfloat162float(bperm[c0 + col]);
#pragma unroll
      for (int i = 0; i < 4; ++i)
        tile[(row + i) * 132 + col] = acc[m][n][i] + b;
    }
  }
  __syncthreads();

  // gate epilogue: item = (row, j-within-block); 4 adjacent tile floats are
  // the r/z/n_i/n_h pre-activations of one (row, j)
  const int j0 = c0 >> 2;  // global j base for this column block (32 j's)
  for (int it = tid; it < BMT * 32; it += 256) {
    const int row = it >> 5;
    const int jj = it & 31;
    const int gr = r0 + row;
    if (gr >= N) continue;
    const f32x4 gse = *reinterpret_cast<const f32x4*>(tile + row * 132 + jj * 4);
    const float r = sigf_(gse[0]);
    const float z = sigf_(gse[1]);
    const float n = tanhf(gse[2] + r * gse[3]);
    const long o = (long)gr * H + j0 + jj;
    const float hv = __bfloat162float(h_in[o]);
    h_new[o] = __float2bfloat16((1.f - z) * n + z * hv);
    R[o] = __float2bfloat16(r);
    Z[o] = __float2bfloat16(z);
    Nn[o] = __float2bfloat16(n);
    HN[o] = __float2bfloat16(gse[3]);
  }
}

void launch_gemm_gru(const bf16* A1, const bf16* A2, const bf16* Wperm,
                     const bf16* bperm, const bf16* h_in, bf16* h_new, bf16* R,
                     bf16* Z, bf16* Nn, bf16* HN, int N, int K, int K1, int H,
                     hipStream_t stream) {
  const int COL = 4 * H;
  const long blocks64 = (long)((N + BM - 1) / BM) * (COL / BN);
  if (blocks64 < 384) {
    const dim3 grid((N + 31) / 32, COL / BN);
    const size_t lds = max((size_t)2 * (32 + BN) * ROWB, (size_t)32 * 132 * 4);
    hipLaunchKernelGGL(gemm_gru_kernel<32>, grid, dim3(256), lds, stream, A1,
                       A2, Wperm, bperm, h_in, h_new, R, Z, Nn, HN, N, K, K1, H);
  } else {
    const dim3 grid((N + BM - 1) / BM, COL / BN);
    const size_t lds = max((size_t)2 * (BM + BN) * ROWB, (size_t)BM * 132 * 4);
    hipLaunchKernelGGL(gemm_gru_kernel<BM>, grid, dim3(256), lds, stream, A1,
                       A2, Wperm, bperm, h_in, h_new, R, Z, Nn, HN, N, K, K1, H);
  }
}
