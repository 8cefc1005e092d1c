// Transformer-shape bf16 GEMM: out(N, COL) = A(N, K) @ W(COL, K)^T [+ bias]
// [+ addend], MI355X (gfx950).
//
// hipBLASLt runs the encoder projections at 173-420 TF on these shapes
// (rocprof, profiles/): this kernel applies the guide's minimum 2-phase
// glds pipeline (cdna_hip_programming.md §5.5 T3 recipe): 128x128 tile,
// BK=64, double-buffered LDS filled by global_load_lds (16-B pieces),
// XOR-swizzle applied on the SOURCE address + the read side (rule 21),
// one vmcnt(0)+barrier per K-tile, s_setprio around the MFMA cluster.
//
// Geometry contract (wrapper-enforced, else fall back to rocBLAS):
// N % 128 == 0, K % 64 == 0, COL % 128 == 0.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define BM2 128
#define BN2 128
#define BK2 64
#define ROWB2 128  // bytes per LDS row (64 bf16)

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ int g2_swz(int row, int byte) {
  return row * ROWB2 + (byte ^ ((row & 7) << 4));
}

// stage one 128x64 bf16 tile into LDS via glds: 16 KiB = 16 pieces of 1 KiB,
// 4 pieces per wave. Lane l of piece c covers (row = 8c + l/8,
// kbyte = (l%8)*16); the source byte offset carries the XOR swizzle.
template <int NP>  // 1-KiB pieces per wave: 4 for 128-row tiles, 2 for 64
__device__ __forceinline__ void g2_stage(const bf16* __restrict__ src_base,
                                         long row_stride_elems, int kk,
                                         char* lds, int wid, int lane) {
#pragma unroll
  for (int c4 = 0; c4 < NP; ++c4) {
    const int c = wid * NP + c4;
    const int row = 8 * c + (lane >> 3);
    const int kbyte = (lane & 7) * 16;
    const int src_byte = kbyte ^ ((row & 7) << 4);
    const bf16* gsrc = src_base + row * row_stride_elems + kk + src_byte / 2;
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int*>(gsrc),
        reinterpret_cast<unsigned int*>(lds + c * 1024 + (lane & 63) * 16), 16, 0, 0);
  }
}

// BMT2 = 128 normally; 64 when the 128-row grid would underfill the chip
// (e.g. the CodeT5 decoder's N=4096 projections: 32x6 = 192 blocks on 512
// resident-block slots)
template <int BMT2>
__global__ __launch_bounds__(256) void gemm2_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ W,
    const float* __restrict__ bias, const bf16* __restrict__ addend,
    bf16* __restrict__ out, int N, int K, int COL) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // double-buffered A and B tiles ((BMT2+128)*128 B each buffer pair)
  // (pointer arrays with addrspace casts are rejected as static
  // initializers; compute the buffer base per use)
#define A_BUF(i) (smem + (i) * (BMT2 * 128 + 16384))
#define B_BUF(i) (smem + BMT2 * 128 + (i) * (BMT2 * 128 + 16384))

  const int r0 = blockIdx.x * BMT2;
  const int c0 = blockIdx.y * BN2;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  const int wm = wid >> 1;  // 2x2 waves, 64x64 tile each
  const int wn = wid & 1;

  constexpr int MF = BMT2 / 32;  // a-fragments per row-wave
  f32x4 acc[MF][4] = {};

  const int nt = K / BK2;
  int cur = 0;
  g2_stage<BMT2 / 32>(A + (long)r0 * K, K, 0, A_BUF(0), wid, lane);
  g2_stage<4>(W + (long)c0 * K, K, 0, B_BUF(0), wid, lane);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < nt; ++t) {
    if (t + 1 < nt) {
      g2_stage<BMT2 / 32>(A + (long)r0 * K, K, (t + 1) * BK2, A_BUF(cur ^ 1), wid, lane);
      g2_stage<4>(W + (long)c0 * K, K, (t + 1) * BK2, B_BUF(cur ^ 1), wid, lane);
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int kbyte = ks * 64 + (lane >> 4) * 16;
      bf16x8 a_frag[MF], b_frag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int brow = wn * 64 + f * 16 + (lane & 15);
        b_frag[f] = *reinterpret_cast<const bf16x8*>(B_BUF(cur) + g2_swz(brow, kbyte));
      }
#pragma unroll
      for (int f = 0; f < MF; ++f) {
        const int arow = wm * (BMT2 / 2) + f * 16 + (lane & 15);
        a_frag[f] = *reinterpret_cast<const bf16x8*>(A_BUF(cur) + g2_swz(arow, kbyte));
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < MF; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: the D fragment layout (col = lane&15, row = (lane>>4)*4+i)
  // would need 64 x 2-B scalar stores per lane (store-issue bound, T21
  // diagnostic); bounce the tile through the now-free LDS and store
  // 16-B-per-lane row-major instead (8 dwordx4 stores per thread).
  __syncthreads();
  char* tile = smem;  // [BMT2][128] bf16
#pragma unroll
  for (int fm = 0; fm < MF; ++fm) {
    const int row = wm * (BMT2 / 2) + fm * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int col = wn * 64 + fn * 16 + (lane & 15);
      const float b = bias ? bias[c0 + col] : 0.f;
#pragma unroll
      for (int i = 0; i < 4; ++i)
        *reinterpret_cast<__bf16*>(tile + (row + i) * 256 + col * 2) =
            (__bf16)(acc[fm][fn][i] + b);
    }
  }
  __syncthreads();
  {
    const int trow = threadIdx.x >> 4;          // 16 rows per pass
    const int tcol = (threadIdx.x & 15) * 8;    // 8 bf16 = 16 B per store
#pragma unroll
    for (int p = 0; p < BMT2 / 16; ++p) {
      const int row = trow + p * 16;
      uint4 v = *reinterpret_cast<const uint4*>(tile + row * 256 + tcol * 2);
      if (addend) {
        __bf16 vals[8];
        *reinterpret_cast<uint4*>(vals) = v;
        const bf16* add = addend + (long)(r0 + row) * COL + c0 + tcol;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vals[j] = (__bf16)((float)vals[j] + __bfloat162float(add[j]));
        v = *reinterpret_cast<const uint4*>(vals);
      }
      *reinterpret_cast<uint4*>(out + (long)(r0 + row) * COL + c0 + tcol) = v;
    }
  }
}

void launch_gemm2(const bf16* A, const bf16* W, const float* bias,
                  const bf16* addend, bf16* out, int N, int K, int COL,
                  hipStream_t stream) {
  const long blocks128 = (long)(N / BM2) * (COL / BN2);
  if (N % 128 == 0 && blocks128 < 512) {
    const dim3 grid(N / 64, COL / BN2);
    hipLaunchKernelGGL(gemm2_kernel<64>, grid, dim3(256), 2 * (64 * 128 + 16384),
                       stream, A, W, bias, addend, out, N, K, COL);
  } else {
    const dim3 grid(N / BM2, COL / BN2);
    hipLaunchKernelGGL(gemm2_kernel<128>, grid, dim3(256), 65536, stream, A, W,
                       bias, addend, out, N, K, COL);
  }
}
