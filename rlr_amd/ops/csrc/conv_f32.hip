// Implicit-GEMM fp32 convolution on MFMA.
// No im2col buffer: the patch matrix is gathered directly into LDS tiles
// and fed to v_mfma_f32_16x16x4_f32 with the same bank padding as
// gemm_f32.hip.
//
//   fwd:        GEMM M=Nb*OH*OW, N=K_out, Kdim=C*R*S (tile 128x64)
//   bwd-data:   GEMM M=Nb*H*W,  N=C,     Kdim=K_out*R*S (tile 128x64),
//               fractional-stride validity masks
//   bwd-weight: GEMM M=K_out,   N=C*R*S, Kdim=Nb*OH*OW (tile 64x64 — Kout
//               is small, a 128-row tile would waste half its M rows),
//               deterministic split-K slabs + fixed-order reduce
//
// The (c,r,s) decomposition in the staging gathers is templated on the
// kernel size (R_T in {1,3}; 0 = generic) and bwd-data on the stride
// (S_T in {1,2}; 0 = generic): constant divisors compile to multiply-shift
// instead of ~20-cycle integer division per staged element — the round-1
// profile showed these gathers dominating (profiles/r01_bench_kernel_stats).
#include "common.h"

typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BK = 32;
constexpr int LDA_S = BK + 2;
constexpr int BN = 64;
constexpr int LDB_S = BN + 16;

struct ConvShape {
  int Nb, C, H, W, Kout, R, S, OH, OW, stride, pad;
};

template <int RT>
__device__ __forceinline__ void crs_decomp(int k, const ConvShape& sh,
                                           int& c, int& r, int& s) {
  if (RT > 0) {
    s = k % RT;
    r = (k / RT) % RT;
    c = k / (RT * RT);
  } else {
    s = k % sh.S;
    r = (k / sh.S) % sh.R;
    c = k / (sh.S * sh.R);
  }
}

// Fragment compute over one staged K-tile.  Wave grid is 2x2; each wave
// owns an (MI*16 x NI*16) output subtile.
template <int MI, int NI>
__device__ __forceinline__ void mfma_tile(const float* __restrict__ Abuf,
                                          const float* __restrict__ Bbuf,
                                          int wr, int wc, int l15, int l4,
                                          f32x4 (&acc)[MI][NI]) {
#pragma unroll
  for (int kk = 0; kk < BK / 4; ++kk) {
    float a_frag[MI], b_frag[NI];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
      a_frag[mi] =
          Abuf[(wr * MI * 16 + mi * 16 + l15) * LDA_S + kk * 4 + l4];
#pragma unroll
    for (int ni = 0; ni < NI; ++ni)
      b_frag[ni] =
          Bbuf[(kk * 4 + l4) * LDB_S + wc * NI * 16 + ni * 16 + l15];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
  }
}

// ------------------------------------------------------------------- fwd

template <int RT, bool P0>
__global__ __launch_bounds__(256)
void conv_fwd_k(const float* __restrict__ x, const float* __restrict__ wt,
                const float* __restrict__ bias, float* __restrict__ y,
                ConvShape sh, int Kdim, int relu) {
  constexpr int BM = 128, MI = 4, NI = 2;
  __shared__ float A_lds[2][BM * LDA_S];
  __shared__ float B_lds[2][BK * LDB_S];
  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int m_blk = blockIdx.x * BM;
  const int n_blk = blockIdx.y * BN;
  const long M = (long)sh.Nb * sh.OH * sh.OW;
  const int am = t >> 3, ak = (t & 7) * 4;
  const int bk = t >> 4, bn = (t & 15) * 4;

  // per-thread m -> (nb, oh, ow) decomposition, once per staging row
  int ows[BM / 32], ohs[BM / 32], nbs[BM / 32];
  bool mval[BM / 32];
#pragma unroll
  for (int j = 0; j < BM / 32; ++j) {
    long gm = m_blk + am + j * 32;
    mval[j] = gm < M;
    long gmc = mval[j] ? gm : 0;
    ows[j] = gmc % sh.OW;
    ohs[j] = (gmc / sh.OW) % sh.OH;
    nbs[j] = gmc / ((long)sh.OW * sh.OH);
  }

  // async-STAGE split (guide T14/G15): loads are issued to REGISTERS a
  // full K-tile early and the LDS write happens after the barrier, so HBM
  // latency hides under the previous tile's MFMAs.
  float ra[BM / 32][4];
  float4 rb[2];
  auto stage_load = [&](int k0) {
#pragma unroll
    for (int j = 0; j < BM / 32; ++j) {
      ra[j][0] = ra[j][1] = ra[j][2] = ra[j][3] = 0.f;
      if (mval[j]) {
        int oh0 = ohs[j] * sh.stride - sh.pad;
        int ow0 = ows[j] * sh.stride - sh.pad;
        const float* xp = x + ((long)nbs[j] * sh.C) * sh.H * sh.W;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          int k = k0 + ak + e;
          if (k < Kdim) {
            int c, r, s;
            crs_decomp<RT>(k, sh, c, r, s);
            int ih = oh0 + r, iw = ow0 + s;
            if (P0 || (ih >= 0 && ih < sh.H && iw >= 0 && iw < sh.W))
              ra[j][e] = xp[((long)c * sh.H + ih) * sh.W + iw];
          }
        }
      }
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int gk = k0 + bk + j * 16;
      float4 q = {0.f, 0.f, 0.f, 0.f};
      if (gk < Kdim) {
        const float* src = wt + (long)gk * sh.Kout + n_blk + bn;
        if (n_blk + bn + 3 < sh.Kout && (sh.Kout % 4) == 0)
          q = *(const float4*)src;
        else {
          if (n_blk + bn + 0 < sh.Kout) q.x = src[0];
          if (n_blk + bn + 1 < sh.Kout) q.y = src[1];
          if (n_blk + bn + 2 < sh.Kout) q.z = src[2];
          if (n_blk + bn + 3 < sh.Kout) q.w = src[3];
        }
      }
      rb[j] = q;
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < BM / 32; ++j) {
      float* dst = &A_lds[buf][(am + j * 32) * LDA_S + ak];
      ((float2*)dst)[0] = {ra[j][0], ra[j][1]};
      ((float2*)dst)[1] = {ra[j][2], ra[j][3]};
    }
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(float4*)&B_lds[buf][(bk + j * 16) * LDB_S + bn] = rb[j];
  };

  stage_load(0);
  stage_write(0);
  if (BK < Kdim) stage_load(BK);
  __syncthreads();
  int buf = 0;
  for (int k0 = 0; k0 < Kdim; k0 += BK) {
    if (k0 + BK < Kdim) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BK < Kdim) stage_load(k0 + 2 * BK);
    }
    mfma_tile<MI, NI>(A_lds[buf], B_lds[buf], wr, wc, l15, l4, acc);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int ko = n_blk + wc * 32 + ni * 16 + l15;
      if (ko >= sh.Kout) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long m = m_blk + wr * 64 + mi * 16 + l4 * 4 + r;
        if (m >= M) continue;
        int ow = m % sh.OW;
        int oh = (m / sh.OW) % sh.OH;
        int nb = m / ((long)sh.OW * sh.OH);
        float v = acc[mi][ni][r];
        if (bias) v += bias[ko];
        if (relu) v = fmaxf(v, 0.f);
        y[(((long)nb * sh.Kout + ko) * sh.OH + oh) * sh.OW + ow] = v;
      }
    }
}

// -------------------------------------------------------------- bwd-data

template <int RT, int ST>
__global__ __launch_bounds__(256)
void conv_bwd_data_k(const float* __restrict__ dy,
                     const float* __restrict__ wp,  // [(ko,r,s)][C]
                     float* __restrict__ dx, ConvShape sh, int Kdim) {
  constexpr int BM = 128, MI = 4, NI = 2;
  __shared__ float A_lds[2][BM * LDA_S];
  __shared__ float B_lds[2][BK * LDB_S];
  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int m_blk = blockIdx.x * BM;
  const int n_blk = blockIdx.y * BN;
  const long M = (long)sh.Nb * sh.H * sh.W;
  const int am = t >> 3, ak = (t & 7) * 4;
  const int bk = t >> 4, bn = (t & 15) * 4;
  const int stride = ST > 0 ? ST : sh.stride;

  int iws[BM / 32], ihs[BM / 32], nbs[BM / 32];
  bool mval[BM / 32];
#pragma unroll
  for (int j = 0; j < BM / 32; ++j) {
    long gm = m_blk + am + j * 32;
    mval[j] = gm < M;
    long gmc = mval[j] ? gm : 0;
    iws[j] = gmc % sh.W;
    ihs[j] = (gmc / sh.W) % sh.H;
    nbs[j] = gmc / ((long)sh.W * sh.H);
  }

  float ra[BM / 32][4];
  float4 rb[2];
  auto stage_load = [&](int k0) {
#pragma unroll
    for (int j = 0; j < BM / 32; ++j) {
      ra[j][0] = ra[j][1] = ra[j][2] = ra[j][3] = 0.f;
      if (mval[j]) {
        int ihp = ihs[j] + sh.pad, iwp = iws[j] + sh.pad;
        const float* dyp =
            dy + ((long)nbs[j] * sh.Kout) * sh.OH * sh.OW;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          int k = k0 + ak + e;
          if (k < Kdim) {
            int ko, r, s;
            crs_decomp<RT>(k, sh, ko, r, s);
            int ohn = ihp - r, own = iwp - s;
            if (ohn >= 0 && own >= 0 && ohn % stride == 0 &&
                own % stride == 0) {
              int oh = ohn / stride, ow = own / stride;
              if (oh < sh.OH && ow < sh.OW)
                ra[j][e] = dyp[((long)ko * sh.OH + oh) * sh.OW + ow];
            }
          }
        }
      }
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int gk = k0 + bk + j * 16;
      float4 q = {0.f, 0.f, 0.f, 0.f};
      if (gk < Kdim) {
        const float* src = wp + (long)gk * sh.C + n_blk + bn;
        if (n_blk + bn + 3 < sh.C && (sh.C % 4) == 0)
          q = *(const float4*)src;
        else {
          if (n_blk + bn + 0 < sh.C) q.x = src[0];
          if (n_blk + bn + 1 < sh.C) q.y = src[1];
          if (n_blk + bn + 2 < sh.C) q.z = src[2];
          if (n_blk + bn + 3 < sh.C) q.w = src[3];
        }
      }
      rb[j] = q;
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < BM / 32; ++j) {
      float* dst = &A_lds[buf][(am + j * 32) * LDA_S + ak];
      ((float2*)dst)[0] = {ra[j][0], ra[j][1]};
      ((float2*)dst)[1] = {ra[j][2], ra[j][3]};
    }
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(float4*)&B_lds[buf][(bk + j * 16) * LDB_S + bn] = rb[j];
  };

  stage_load(0);
  stage_write(0);
  if (BK < Kdim) stage_load(BK);
  __syncthreads();
  int buf = 0;
  for (int k0 = 0; k0 < Kdim; k0 += BK) {
    if (k0 + BK < Kdim) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BK < Kdim) stage_load(k0 + 2 * BK);
    }
    mfma_tile<MI, NI>(A_lds[buf], B_lds[buf], wr, wc, l15, l4, acc);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int c = n_blk + wc * 32 + ni * 16 + l15;
      if (c >= sh.C) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long m = m_blk + wr * 64 + mi * 16 + l4 * 4 + r;
        if (m >= M) continue;
        int iw = m % sh.W;
        int ih = (m / sh.W) % sh.H;
        int nb = m / ((long)sh.W * sh.H);
        dx[(((long)nb * sh.C + c) * sh.H + ih) * sh.W + iw] =
            acc[mi][ni][r];
      }
    }
}

// ------------------------------------------------------------ bwd-weight

// 64x64 tile (MI=NI=2): Kout rarely exceeds 64 per tile row and a 128-row
// tile would idle half its MFMAs.  gridDim.z = split-K chunks over
// m = (nb,oh,ow); partial slabs [z][Kout][C*R*S].
template <int RT, bool P0>
__global__ __launch_bounds__(256)
void conv_bwd_weight_k(const float* __restrict__ dy,
                       const float* __restrict__ x, float* __restrict__ out,
                       ConvShape sh, int Ncrs, long k_per_chunk,
                       int direct_out) {
  constexpr int BM = 64, MI = 2, NI = 2;
  __shared__ float A_lds[2][BM * LDA_S];
  __shared__ float B_lds[2][BK * LDB_S];
  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int m_blk = blockIdx.x * BM;   // over Kout
  const int n_blk = blockIdx.y * BN;   // over C*R*S
  const long Kdim = (long)sh.Nb * sh.OH * sh.OW;
  const long k_lo = (long)blockIdx.z * k_per_chunk;
  const long k_hi = min(Kdim, k_lo + k_per_chunk);
  // A staging: BM*BK = 2048 floats / 256 threads = 8 = 2 x float4
  const int am = t >> 3, ak = (t & 7) * 4;    // am 0..31, 2 rounds
  const int bk = t >> 4, bn = (t & 15) * 4;

  // precompute the (c,r,s) for this thread's 4 B columns (fixed all tiles)
  int bc[4], br[4], bs[4];
  bool bvalid[4];
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    int crs = n_blk + bn + e;
    bvalid[e] = crs < Ncrs;
    crs_decomp<RT>(bvalid[e] ? crs : 0, sh, bc[e], br[e], bs[e]);
  }

  float ra[2][4];
  float4 rb[2];
  auto stage_load = [&](long k0) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int ko = m_blk + am + j * 32;
      ra[j][0] = ra[j][1] = ra[j][2] = ra[j][3] = 0.f;
      if (ko < sh.Kout) {
        // 4 consecutive m share (nb, oh) almost always; slow path on wrap
        long k = k0 + ak;
        int ow = k % sh.OW;
        int oh = (k / sh.OW) % sh.OH;
        int nb = k / ((long)sh.OW * sh.OH);
        const float* dyp =
            dy + (((long)nb * sh.Kout + ko) * sh.OH + oh) * sh.OW;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          if (k + e < k_hi) {
            int owe = ow + e;
            if (owe < sh.OW)
              ra[j][e] = dyp[owe];
            else {
              long ke = k + e;
              int ow2 = ke % sh.OW;
              int oh2 = (ke / sh.OW) % sh.OH;
              int nb2 = ke / ((long)sh.OW * sh.OH);
              ra[j][e] = dy[(((long)nb2 * sh.Kout + ko) * sh.OH + oh2) *
                                sh.OW + ow2];
            }
          }
        }
      }
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      long k = k0 + bk + j * 16;
      float v[4] = {0.f, 0.f, 0.f, 0.f};
      if (k < k_hi) {
        int ow = k % sh.OW;
        int oh = (k / sh.OW) % sh.OH;
        int nb = k / ((long)sh.OW * sh.OH);
        int ih0 = oh * sh.stride - sh.pad;
        int iw0 = ow * sh.stride - sh.pad;
        const float* xp = x + ((long)nb * sh.C) * sh.H * sh.W;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          if (bvalid[e]) {
            int ih = ih0 + br[e], iw = iw0 + bs[e];
            if (P0 || (ih >= 0 && ih < sh.H && iw >= 0 && iw < sh.W))
              v[e] = xp[((long)bc[e] * sh.H + ih) * sh.W + iw];
          }
        }
      }
      rb[j] = {v[0], v[1], v[2], v[3]};
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      float* dst = &A_lds[buf][(am + j * 32) * LDA_S + ak];
      ((float2*)dst)[0] = {ra[j][0], ra[j][1]};
      ((float2*)dst)[1] = {ra[j][2], ra[j][3]};
    }
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(float4*)&B_lds[buf][(bk + j * 16) * LDB_S + bn] = rb[j];
  };

  stage_load(k_lo);
  stage_write(0);
  if (k_lo + BK < k_hi) stage_load(k_lo + BK);
  __syncthreads();
  int buf = 0;
  for (long k0 = k_lo; k0 < k_hi; k0 += BK) {
    if (k0 + BK < k_hi) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BK < k_hi) stage_load(k0 + 2 * BK);
    }
    mfma_tile<MI, NI>(A_lds[buf], B_lds[buf], wr, wc, l15, l4, acc);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int crs = n_blk + wc * 32 + ni * 16 + l15;
      if (crs >= Ncrs) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int ko = m_blk + wr * 32 + mi * 16 + l4 * 4 + r;
        if (ko >= sh.Kout) continue;
        if (direct_out)
          out[(long)ko * Ncrs + crs] = acc[mi][ni][r];
        else
          out[((long)blockIdx.z * sh.Kout + ko) * Ncrs + crs] =
              acc[mi][ni][r];
      }
    }
}

// ---- conv bias gradient: db[k] = sum over (nb,oh,ow) of dy ----
// Two-stage deterministic reduce: stage 1 fills partials[k][chunk] from a
// (Kout x NCHUNK) grid (plenty of blocks for 256 CUs); stage 2 is one wave
// per k over the NCHUNK partials.
constexpr int kDbChunks = 64;

__global__ void conv_db_stage1_k(const float* __restrict__ dy,
                                 float* __restrict__ partials, int Nb,
                                 int Kout, int OHW) {
  int k = blockIdx.x;
  int chunk = blockIdx.y;
  long total = (long)Nb * OHW;
  long per = (total + kDbChunks - 1) / kDbChunks;
  long lo = chunk * per, hi = min(total, lo + per);
  __shared__ float sh[kBlock];
  float acc = 0.f;
  for (long i = lo + threadIdx.x; i < hi; i += blockDim.x) {
    long nb = i / OHW, px = i % OHW;
    acc += dy[(nb * Kout + k) * (long)OHW + px];
  }
  sh[threadIdx.x] = acc;
  __syncthreads();
  for (int off = kBlock / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sh[threadIdx.x] += sh[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) partials[(long)k * kDbChunks + chunk] = sh[0];
}

__global__ void conv_db_stage2_k(const float* __restrict__ partials,
                                 float* __restrict__ db, int Kout) {
  int k = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  int lane = threadIdx.x % kWave;
  if (k >= Kout) return;
  float acc = (lane < kDbChunks) ? partials[(long)k * kDbChunks + lane] : 0.f;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, kWave);
  if (lane == 0) db[k] = acc;
}

// permute w (Kout,C,R,S) -> dst layouts
__global__ void wperm_crs_ko_k(const float* __restrict__ w,
                               float* __restrict__ out, int Kout, int C,
                               int RS) {
  long n = (long)Kout * C * RS;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int ko = i / (C * RS);
    int crs = i % (C * RS);
    out[(long)crs * Kout + ko] = w[i];
  }
}

__global__ void wperm_kors_c_k(const float* __restrict__ w,
                               float* __restrict__ out, int Kout, int C,
                               int RS) {
  long n = (long)Kout * C * RS;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int rs = i % RS;
    int c = (i / RS) % C;
    int ko = i / ((long)RS * C);
    out[((long)ko * RS + rs) * C + c] = w[i];
  }
}

extern "C" void launch_splitk_reduce(const float* ws, float* C,
                                     const float* bias, int M, int N,
                                     int ldc, int SK, int relu, void* s);

extern "C" {
void launch_conv_fwd(const float* x, const float* wt, const float* bias,
                     float* y, int Nb, int C, int H, int W, int Kout, int R,
                     int S, int OH, int OW, int stride, int pad, int relu,
                     void* s) {
  ConvShape sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Kdim = C * R * S;
  long M = (long)Nb * OH * OW;
  dim3 grid((M + 127) / 128, (Kout + BN - 1) / BN, 1);
  hipStream_t st = (hipStream_t)s;
  if (R == 3 && S == 3 && pad == 0)
    conv_fwd_k<3, true><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim, relu);
  else if (R == 3 && S == 3)
    conv_fwd_k<3, false><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim, relu);
  else if (R == 1 && S == 1 && pad == 0)
    conv_fwd_k<1, true><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim, relu);
  else
    conv_fwd_k<0, false><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim, relu);
}

void launch_conv_bwd_data(const float* dy, const float* wp, float* dx,
                          int Nb, int C, int H, int W, int Kout, int R,
                          int S, int OH, int OW, int stride, int pad,
                          void* s) {
  ConvShape sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Kdim = Kout * R * S;
  long M = (long)Nb * H * W;
  dim3 grid((M + 127) / 128, (C + BN - 1) / BN, 1);
  hipStream_t st = (hipStream_t)s;
  if (R == 3 && S == 3 && stride == 1)
    conv_bwd_data_k<3, 1><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim);
  else if (R == 3 && S == 3 && stride == 2)
    conv_bwd_data_k<3, 2><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim);
  else if (R == 1 && S == 1 && stride == 2)
    conv_bwd_data_k<1, 2><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim);
  else if (R == 1 && S == 1 && stride == 1)
    conv_bwd_data_k<1, 1><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim);
  else
    conv_bwd_data_k<0, 0><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim);
}

int conv_bwd_weight_splitk(int Kout, int Ncrs, long Kdim) {
  long tiles = ((Kout + 63) / 64) * (long)((Ncrs + BN - 1) / BN);
  if (tiles >= 192 || Kdim <= 2 * BK) return 1;
  long want = (512 + tiles - 1) / tiles;  // 2 blocks/CU: latency hiding
  long max_chunks = (Kdim + BK - 1) / BK;
  long sk = want < max_chunks ? want : max_chunks;
  return (int)(sk < 1 ? 1 : (sk > 256 ? 256 : sk));
}

void launch_conv_bwd_weight(const float* dy, const float* x, float* dw,
                            float* ws, int SK, int Nb, int C, int H, int W,
                            int Kout, int R, int S, int OH, int OW,
                            int stride, int pad, void* s) {
  ConvShape sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Ncrs = C * R * S;
  long Kdim = (long)Nb * OH * OW;
  long k_per_chunk =
      SK == 1 ? Kdim : (((Kdim + SK - 1) / SK + BK - 1) / BK) * BK;
  dim3 grid((Kout + 63) / 64, (Ncrs + BN - 1) / BN, SK);
  hipStream_t st = (hipStream_t)s;
  float* out = SK == 1 ? dw : ws;
  if (R == 3 && S == 3 && pad == 0)
    conv_bwd_weight_k<3, true><<<grid, 256, 0, st>>>(dy, x, out, sh, Ncrs,
                                                     k_per_chunk, SK == 1);
  else if (R == 3 && S == 3)
    conv_bwd_weight_k<3, false><<<grid, 256, 0, st>>>(dy, x, out, sh, Ncrs,
                                                      k_per_chunk, SK == 1);
  else if (R == 1 && S == 1 && pad == 0)
    conv_bwd_weight_k<1, true><<<grid, 256, 0, st>>>(dy, x, out, sh, Ncrs,
                                                     k_per_chunk, SK == 1);
  else
    conv_bwd_weight_k<0, false><<<grid, 256, 0, st>>>(dy, x, out, sh, Ncrs,
                                                      k_per_chunk, SK == 1);
  if (SK > 1)
    launch_splitk_reduce(ws, dw, nullptr, Kout, Ncrs, Ncrs, SK, 0, s);
}

void launch_conv_db(const float* dy, float* db, float* partials, int Nb,
                    int Kout, int OHW, void* s) {
  hipStream_t st = (hipStream_t)s;
  conv_db_stage1_k<<<dim3(Kout, kDbChunks), kBlock, 0, st>>>(dy, partials,
                                                             Nb, Kout, OHW);
  int wpb = kBlock / kWave;
  conv_db_stage2_k<<<(Kout + wpb - 1) / wpb, kBlock, 0, st>>>(partials, db,
                                                              Kout);
}

void launch_wperm_crs_ko(const float* w, float* out, int Kout, int C, int RS,
                         void* s) {
  wperm_crs_ko_k<<<grid_for((long)Kout * C * RS), kBlock, 0,
                   (hipStream_t)s>>>(w, out, Kout, C, RS);
}
void launch_wperm_kors_c(const float* w, float* out, int Kout, int C, int RS,
                         void* s) {
  wperm_kors_c_k<<<grid_for((long)Kout * C * RS), kBlock, 0,
                   (hipStream_t)s>>>(w, out, Kout, C, RS);
}
}
