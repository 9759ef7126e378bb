// Implicit-GEMM fp32 convolution on MFMA.
// No im2col buffer: the patch matrix is gathered directly into LDS tiles
// and fed to v_mfma_f32_16x16x4_f32 — same 128x64x32 tile structure and
// bank padding as gemm_f32.hip.
//
//   fwd:        y[(nb,ko,oh,ow)] = sum_{c,r,s} x[nb,c,ihw] * w[ko,c,r,s]
//               GEMM M=Nb*OH*OW, N=K_out, Kdim=C*R*S; B = w permuted to
//               [(c,r,s)][ko] (tiny transform, done by the binding)
//   bwd-data:   GEMM M=Nb*H*W, N=C, Kdim=K_out*R*S over dy with
//               fractional-stride validity masks; B = w permuted to
//               [(ko,r,s)][c]
//   bwd-weight: GEMM M=K_out, N=C*R*S, Kdim=Nb*OH*OW, deterministic
//               split-K slabs + fixed-order reduce (no atomics)
//
// General stride/pad (the reference needs only stride1/pad0,
// models.py:14-38; ResNet18 adds stride 2 and pad 1).
#include "common.h"

typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BM = 128, BN = 64, BK = 32;
constexpr int LDA_S = BK + 2;
constexpr int LDB_S = BN + 16;

struct ConvShape {
  int Nb, C, H, W, Kout, R, S, OH, OW, stride, pad;
};

// ---------------------------------------------------------------- helpers

#define MFMA_CORE()                                                          \
  const int t = threadIdx.x;                                                 \
  const int wave = t >> 6, lane = t & 63;                                    \
  const int wr = wave >> 1, wc = wave & 1;                                   \
  const int l15 = lane & 15, l4 = lane >> 4;                                 \
  f32x4 acc[4][2];                                                           \
  _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                           \
      _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                       \
          acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

#define MFMA_TILE(Abuf, Bbuf)                                                \
  _Pragma("unroll") for (int kk = 0; kk < BK / 4; ++kk) {                    \
    float a_frag[4], b_frag[2];                                              \
    _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                         \
        a_frag[mi] = Abuf[(wr * 64 + mi * 16 + l15) * LDA_S + kk * 4 + l4];  \
    _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                         \
        b_frag[ni] = Bbuf[(kk * 4 + l4) * LDB_S + wc * 32 + ni * 16 + l15];  \
    _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                         \
        _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                     \
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(              \
                a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);               \
  }

// ------------------------------------------------------------------- fwd

__global__ __launch_bounds__(256)
void conv_fwd_k(const float* __restrict__ x, const float* __restrict__ wt,
                const float* __restrict__ bias, float* __restrict__ y,
                ConvShape sh, int Kdim, int relu) {
  __shared__ float A_lds[2][BM * LDA_S];
  __shared__ float B_lds[2][BK * LDB_S];
  MFMA_CORE();

  const int m_blk = blockIdx.x * BM;
  const int n_blk = blockIdx.y * BN;
  const long M = (long)sh.Nb * sh.OH * sh.OW;
  const int am = t >> 3, ak = (t & 7) * 4;
  const int bk = t >> 4, bn = (t & 15) * 4;

  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int m = am + j * 32;
      long gm = m_blk + m;
      float v[4] = {0.f, 0.f, 0.f, 0.f};
      if (gm < M) {
        int ow = gm % sh.OW;
        int oh = (gm / sh.OW) % sh.OH;
        int nb = gm / ((long)sh.OW * sh.OH);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          int k = k0 + ak + e;
          if (k < Kdim) {
            int s = k % sh.S;
            int r = (k / sh.S) % sh.R;
            int c = k / (sh.S * sh.R);
            int ih = oh * sh.stride - sh.pad + r;
            int iw = ow * sh.stride - sh.pad + s;
            if (ih >= 0 && ih < sh.H && iw >= 0 && iw < sh.W)
              v[e] = x[(((long)nb * sh.C + c) * sh.H + ih) * sh.W + iw];
          }
        }
      }
      float* dst = &A_lds[buf][m * LDA_S + ak];
      ((float2*)dst)[0] = {v[0], v[1]};
      ((float2*)dst)[1] = {v[2], v[3]};
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int kk = bk + j * 16;
      int gk = k0 + kk;
      float4 q = {0.f, 0.f, 0.f, 0.f};
      if (gk < Kdim) {
        // wt: [(c,r,s)][Kout], rows 16B-aligned iff Kout%4==0
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
      *(float4*)&B_lds[buf][kk * LDB_S + bn] = q;
    }
  };

  stage(0, 0);
  __syncthreads();
  int buf = 0;
  for (int k0 = 0; k0 < Kdim; k0 += BK) {
    if (k0 + BK < Kdim) stage(buf ^ 1, k0 + BK);
    MFMA_TILE(A_lds[buf], B_lds[buf]);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
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

__global__ __launch_bounds__(256)
void conv_bwd_data_k(const float* __restrict__ dy,
                     const float* __restrict__ wp,  // [(ko,r,s)][C]
                     float* __restrict__ dx, ConvShape sh, int Kdim) {
  __shared__ float A_lds[2][BM * LDA_S];
  __shared__ float B_lds[2][BK * LDB_S];
  MFMA_CORE();

  const int m_blk = blockIdx.x * BM;
  const int n_blk = blockIdx.y * BN;
  const long M = (long)sh.Nb * sh.H * sh.W;
  const int am = t >> 3, ak = (t & 7) * 4;
  const int bk = t >> 4, bn = (t & 15) * 4;

  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int m = am + j * 32;
      long gm = m_blk + m;
      float v[4] = {0.f, 0.f, 0.f, 0.f};
      if (gm < M) {
        int iw = gm % sh.W;
        int ih = (gm / sh.W) % sh.H;
        int nb = gm / ((long)sh.W * sh.H);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          int k = k0 + ak + e;
          if (k < Kdim) {
            int s = k % sh.S;
            int r = (k / sh.S) % sh.R;
            int ko = k / (sh.S * sh.R);
            int ohn = ih + sh.pad - r;
            int own = iw + sh.pad - s;
            if (ohn >= 0 && own >= 0 && ohn % sh.stride == 0 &&
                own % sh.stride == 0) {
              int oh = ohn / sh.stride, ow = own / sh.stride;
              if (oh < sh.OH && ow < sh.OW)
                v[e] = dy[(((long)nb * sh.Kout + ko) * sh.OH + oh) * sh.OW +
                          ow];
            }
          }
        }
      }
      float* dst = &A_lds[buf][m * LDA_S + ak];
      ((float2*)dst)[0] = {v[0], v[1]};
      ((float2*)dst)[1] = {v[2], v[3]};
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int kk = bk + j * 16;
      int gk = k0 + kk;
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
      *(float4*)&B_lds[buf][kk * LDB_S + bn] = q;
    }
  };

  stage(0, 0);
  __syncthreads();
  int buf = 0;
  for (int k0 = 0; k0 < Kdim; k0 += BK) {
    if (k0 + BK < Kdim) stage(buf ^ 1, k0 + BK);
    MFMA_TILE(A_lds[buf], B_lds[buf]);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
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

// A[ko][m=(nb,oh,ow)] = dy gather; B[m][(c,r,s)] = x patch gather.
// gridDim.z = split-K chunks over m; partial slabs [z][Kout][C*R*S].
__global__ __launch_bounds__(256)
void conv_bwd_weight_k(const float* __restrict__ dy,
                       const float* __restrict__ x, float* __restrict__ out,
                       ConvShape sh, int Ncrs, long k_per_chunk,
                       int direct_out) {
  __shared__ float A_lds[2][BM * LDA_S];
  __shared__ float B_lds[2][BK * LDB_S];
  MFMA_CORE();

  const int m_blk = blockIdx.x * BM;   // over Kout
  const int n_blk = blockIdx.y * BN;   // over C*R*S
  const long Kdim = (long)sh.Nb * sh.OH * sh.OW;
  const long k_lo = (long)blockIdx.z * k_per_chunk;
  const long k_hi = min(Kdim, k_lo + k_per_chunk);
  const int am = t >> 3, ak = (t & 7) * 4;
  const int bk = t >> 4, bn = (t & 15) * 4;

  auto stage = [&](int buf, long k0) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int ko = m_blk + am + j * 32;
      float v[4] = {0.f, 0.f, 0.f, 0.f};
      if (ko < sh.Kout) {
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          long k = k0 + ak + e;
          if (k < k_hi) {
            int ow = k % sh.OW;
            int oh = (k / sh.OW) % sh.OH;
            int nb = k / ((long)sh.OW * sh.OH);
            v[e] = dy[(((long)nb * sh.Kout + ko) * sh.OH + oh) * sh.OW + ow];
          }
        }
      }
      float* dst = &A_lds[buf][(am + j * 32) * LDA_S + ak];
      ((float2*)dst)[0] = {v[0], v[1]};
      ((float2*)dst)[1] = {v[2], v[3]};
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int kk = bk + j * 16;
      long k = k0 + kk;
      float v[4] = {0.f, 0.f, 0.f, 0.f};
      if (k < k_hi) {
        int ow = k % sh.OW;
        int oh = (k / sh.OW) % sh.OH;
        int nb = k / ((long)sh.OW * sh.OH);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          int crs = n_blk + bn + e;
          if (crs < Ncrs) {
            int s = crs % sh.S;
            int r = (crs / sh.S) % sh.R;
            int c = crs / (sh.S * sh.R);
            int ih = oh * sh.stride - sh.pad + r;
            int iw = ow * sh.stride - sh.pad + s;
            if (ih >= 0 && ih < sh.H && iw >= 0 && iw < sh.W)
              v[e] = x[(((long)nb * sh.C + c) * sh.H + ih) * sh.W + iw];
          }
        }
      }
      *(float4*)&B_lds[buf][kk * LDB_S + bn] = {v[0], v[1], v[2], v[3]};
    }
  };

  stage(0, k_lo);
  __syncthreads();
  int buf = 0;
  for (long k0 = k_lo; k0 < k_hi; k0 += BK) {
    if (k0 + BK < k_hi) stage(buf ^ 1, k0 + BK);
    MFMA_TILE(A_lds[buf], B_lds[buf]);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      int crs = n_blk + wc * 32 + ni * 16 + l15;
      if (crs >= Ncrs) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int ko = m_blk + wr * 64 + mi * 16 + l4 * 4 + r;
        if (ko >= sh.Kout) continue;
        if (direct_out)
          out[(long)ko * Ncrs + crs] = acc[mi][ni][r];
        else
          out[((long)blockIdx.z * sh.Kout + ko) * Ncrs + crs] =
              acc[mi][ni][r];
      }
    }
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
    // i = ((ko*C + c)*RS + rs)  ->  out[((ko*RS + rs)*C + c)]
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
  dim3 grid((M + BM - 1) / BM, (Kout + BN - 1) / BN, 1);
  conv_fwd_k<<<grid, 256, 0, (hipStream_t)s>>>(x, wt, bias, y, sh, Kdim,
                                               relu);
}

void launch_conv_bwd_data(const float* dy, const float* wp, float* dx,
                          int Nb, int C, int H, int W, int Kout, int R,
                          int S, int OH, int OW, int stride, int pad,
                          void* s) {
  ConvShape sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Kdim = Kout * R * S;
  long M = (long)Nb * H * W;
  dim3 grid((M + BM - 1) / BM, (C + BN - 1) / BN, 1);
  conv_bwd_data_k<<<grid, 256, 0, (hipStream_t)s>>>(dy, wp, dx, sh, Kdim);
}

int conv_bwd_weight_splitk(int Kout, int Ncrs, long Kdim) {
  long tiles = ((Kout + BM - 1) / BM) * (long)((Ncrs + BN - 1) / BN);
  if (tiles >= 192 || Kdim <= 2 * BK) return 1;
  long want = (256 + tiles - 1) / tiles;
  long max_chunks = (Kdim + BK - 1) / BK;
  long sk = want < max_chunks ? want : max_chunks;
  return (int)(sk < 1 ? 1 : (sk > 128 ? 128 : sk));
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
  dim3 grid((Kout + BM - 1) / BM, (Ncrs + BN - 1) / BN, SK);
  hipStream_t st = (hipStream_t)s;
  conv_bwd_weight_k<<<grid, 256, 0, st>>>(dy, x, SK == 1 ? dw : ws, sh, Ncrs,
                                          k_per_chunk, SK == 1);
  if (SK > 1)
    launch_splitk_reduce(ws, dw, nullptr, Kout, Ncrs, Ncrs, SK, 0, s);
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
