// MFMA bf16 GEMM for gfx950 — the reduced-precision path for the build's
// ResNet18/CIFAR configs (BASELINE.json config 3; --dtype bf16).
//
// C[M,N](f32 or bf16) = A[M,K](bf16) x B[K,N](bf16), fp32 accumulate via
// v_mfma_f32_16x16x32_bf16 (8 bf16 per lane per operand, f32x4
// accumulator; cdna_hip_programming.md §3).  Fragment A: lane l holds
// A[row = l&15][k = (l>>4)*8 .. +7]; B: B[k = (l>>4)*8+j][col = l&15];
// C/D: col = l&15, row = (l>>4)*4 + reg (verified against torch matmul in
// tests/test_bf16_gpu.py with asymmetric operands — guide G9).
//
// Structure mirrors gemm_f32.hip: 256-thread block (2x2 waves), tile
// 128x64, K-step 32 (one MFMA k-depth), double-buffered LDS with the
// async-stage split.  LDS layouts:
//   A_lds[BM][BK]   bf16, row-padded +8 elements (16 B) — lane groups of
//                   a b128 read hit distinct bank pairs
//   B_lds[BK][BN+8] bf16 — B fragment reads are 16-byte per lane along k,
//                   so B is staged TRANSPOSED as [n][k] for contiguity:
//                   B_lds[n][k = 0..31]
#include "common.h"
#include <hip/hip_bf16.h>

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef short bf16x8 __attribute__((ext_vector_type(8)));

constexpr int BM16 = 128, BN16 = 64, BK16 = 32;
constexpr int LDA16 = BK16 + 8;   // bf16 elements per A_lds row
constexpr int LDB16 = BK16 + 8;   // bf16 elements per B_lds row (B^T image)

// A and B both staged as [row][k] bf16 (B transposed at stage time), so a
// fragment read is ONE 16-byte load of 8 contiguous bf16.
template <bool VEC>
__global__ __launch_bounds__(256)
void gemm_bf16_k(const unsigned short* __restrict__ A,
                 const unsigned short* __restrict__ B,  // [K][N] row-major
                 float* __restrict__ C, const float* __restrict__ bias,
                 unsigned short* __restrict__ C16, int M, int N, int K,
                 int lda, int ldb, int ldc, long k_per_chunk, int relu,
                 int direct_out) {
  __shared__ unsigned short A_lds[2][BM16 * LDA16];
  __shared__ unsigned short B_lds[2][BN16 * LDB16];

  const int m_blk = blockIdx.x * BM16;
  const int n_blk = blockIdx.y * BN16;
  const long k_lo = (long)blockIdx.z * k_per_chunk;
  const long k_hi = min((long)K, k_lo + k_per_chunk);

  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;

  f32x4 acc[4][2];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  // A staging: 128 rows x 32 k bf16 = 8 KB; 256 threads x 2 rounds of
  // 8 bf16 (16 B).  thread t: row = t>>2 (+64/round), k = (t&3)*8.
  const int am = t >> 2, ak = (t & 3) * 8;
  // B staging: read B [K][N] rows (n contiguous), write TRANSPOSED image
  // B_lds[n][k].  thread t: k-row = t>>3 (+... 32 rows), n = (t&7)*8.
  const int bkr = t >> 3, bnc = (t & 7) * 8;

  unsigned short ra[2][8];
  unsigned short rbv[8];
  auto stage_load = [&](long k0) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      long gm = m_blk + am + j * 64;
      long gk = k0 + ak;
#pragma unroll
      for (int e = 0; e < 8; ++e) ra[j][e] = 0;
      if (gm < M) {
        if (VEC && gk + 7 < k_hi) {
          *(bf16x8*)ra[j] =
              *(const bf16x8*)(A + gm * (long)lda + gk);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (gk + e < k_hi) ra[j][e] = A[gm * (long)lda + gk + e];
        }
      }
    }
    {
      long gk = k0 + bkr;
#pragma unroll
      for (int e = 0; e < 8; ++e) rbv[e] = 0;
      if (gk < k_hi) {
        if (VEC && n_blk + bnc + 7 < N) {
          *(bf16x8*)rbv = *(const bf16x8*)(B + gk * (long)ldb + n_blk + bnc);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (n_blk + bnc + e < N)
              rbv[e] = B[gk * (long)ldb + n_blk + bnc + e];
        }
      }
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(bf16x8*)&A_lds[buf][(am + j * 64) * LDA16 + ak] = *(bf16x8*)ra[j];
    // transpose B: element (k = bkr, n = bnc+e) -> B_lds[n][k]
#pragma unroll
    for (int e = 0; e < 8; ++e)
      B_lds[buf][(bnc + e) * LDB16 + bkr] = rbv[e];
  };

  stage_load(k_lo);
  stage_write(0);
  if (k_lo + BK16 < k_hi) stage_load(k_lo + BK16);
  __syncthreads();

  int buf = 0;
  for (long k0 = k_lo; k0 < k_hi; k0 += BK16) {
    if (k0 + BK16 < k_hi) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BK16 < k_hi) stage_load(k0 + 2 * BK16);
    }
    const unsigned short* Ab = A_lds[buf];
    const unsigned short* Bb = B_lds[buf];
    // one 16x16x32 MFMA covers the whole BK16: fragment k = l4*8..l4*8+7
    bf16x8 a_frag[4], b_frag[2];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      a_frag[mi] = *(const bf16x8*)&Ab[(wr * 64 + mi * 16 + l15) * LDA16 +
                                      l4 * 8];
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
      b_frag[ni] = *(const bf16x8*)&Bb[(wc * 32 + ni * 16 + l15) * LDB16 +
                                       l4 * 8];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      int col = n_blk + wc * 32 + ni * 16 + l15;
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m_blk + wr * 64 + mi * 16 + l4 * 4 + r;
        if (row >= M) continue;
        float v = acc[mi][ni][r];
        if (direct_out) {
          if (bias) v += bias[col];
          if (relu) v = fmaxf(v, 0.f);
          if (C16)
            C16[(long)row * ldc + col] = f2bf_(v);
          else
            C[(long)row * ldc + col] = v;
        } else {
          C[((long)blockIdx.z * M + row) * N + col] = v;
        }
      }
    }
}

// f32 <-> bf16 casts (flat)
__global__ void f32_to_bf16_k(const float* __restrict__ in,
                              unsigned short* __restrict__ out, long n) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = f2bf_(in[i]);
}

__global__ void bf16_to_f32_k(const unsigned short* __restrict__ in,
                              float* __restrict__ out, long n) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = bf2f_(in[i]);
}

extern "C" {
void launch_splitk_reduce(const float* ws, float* C, const float* bias,
                          int M, int N, int ldc, int SK, int relu, void* s);
int gemm_f32_splitk(int M, int N, int K);

// C is fp32 when c16 == nullptr, bf16 otherwise (split-K always reduces in
// fp32 then the caller casts if needed — split-K path requires c16==null).
void launch_gemm_bf16(const unsigned short* A, const unsigned short* B,
                      float* C, const float* bias, unsigned short* C16,
                      float* ws, int M, int N, int K, int lda, int ldb,
                      int ldc, int SK, int relu, void* s) {
  hipStream_t st = (hipStream_t)s;
  dim3 grid((M + BM16 - 1) / BM16, (N + BN16 - 1) / BN16, SK);
  long k_per_chunk =
      SK == 1 ? (long)K
              : ((((long)K + SK - 1) / SK + BK16 - 1) / BK16) * BK16;
  bool vec = (lda % 8 == 0) && (ldb % 8 == 0);
  float* out = SK == 1 ? C : ws;
  if (vec)
    gemm_bf16_k<true><<<grid, 256, 0, st>>>(A, B, out, bias,
                                            SK == 1 ? C16 : nullptr, M, N,
                                            K, lda, ldb, ldc, k_per_chunk,
                                            relu, SK == 1);
  else
    gemm_bf16_k<false><<<grid, 256, 0, st>>>(A, B, out, bias,
                                             SK == 1 ? C16 : nullptr, M, N,
                                             K, lda, ldb, ldc, k_per_chunk,
                                             relu, SK == 1);
  if (SK > 1) launch_splitk_reduce(ws, C, bias, M, N, ldc, SK, relu, s);
}

void launch_f32_to_bf16(const float* in, unsigned short* out, long n,
                        void* s) {
  f32_to_bf16_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(in, out, n);
}
void launch_bf16_to_f32(const unsigned short* in, float* out, long n,
                        void* s) {
  bf16_to_f32_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(in, out, n);
}
}
