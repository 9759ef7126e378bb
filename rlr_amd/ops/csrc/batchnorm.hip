// BatchNorm2d fwd/bwd (build extension for ResNet18 — the reference has no
// BN anywhere, SURVEY.md §2b last row).  NCHW, per-channel statistics.
// Deterministic: one block per channel, fixed-order tree reductions, no
// atomics.
#include "common.h"

// ---- stage 1 (train fwd): per-channel sum and sum-of-squares ----
__global__ void bn_stats_k(const float* __restrict__ x, float* __restrict__
                           sums,  // [2][C]
                           int Nb, int C, int HW) {
  int c = blockIdx.x;
  __shared__ float sh[2][kBlock];
  float s = 0.f, ss = 0.f;
  for (long i = threadIdx.x; i < (long)Nb * HW; i += blockDim.x) {
    long nb = i / HW, px = i % HW;
    float v = x[(nb * C + c) * (long)HW + px];
    s += v;
    ss += v * v;
  }
  sh[0][threadIdx.x] = s;
  sh[1][threadIdx.x] = ss;
  __syncthreads();
  for (int off = kBlock / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      sh[0][threadIdx.x] += sh[0][threadIdx.x + off];
      sh[1][threadIdx.x] += sh[1][threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    sums[c] = sh[0][0];
    sums[C + c] = sh[1][0];
  }
}

// ---- stage 2: finalize mean/rstd, update running stats (torch semantics:
// running_var uses the UNBIASED batch variance) ----
__global__ void bn_finalize_k(const float* __restrict__ sums,
                              float* __restrict__ save_mean,
                              float* __restrict__ save_rstd,
                              float* __restrict__ running_mean,
                              float* __restrict__ running_var, int C,
                              long count, float momentum, float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = sums[c] / count;
  float var = sums[C + c] / count - mean * mean;
  var = fmaxf(var, 0.f);
  save_mean[c] = mean;
  save_rstd[c] = rsqrtf(var + eps);
  if (running_mean) {
    float unbiased = count > 1 ? var * count / (count - 1) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// eval path: mean/rstd from running stats
__global__ void bn_eval_stats_k(const float* __restrict__ running_mean,
                                const float* __restrict__ running_var,
                                float* __restrict__ save_mean,
                                float* __restrict__ save_rstd, int C,
                                float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  save_mean[c] = running_mean[c];
  save_rstd[c] = rsqrtf(running_var[c] + eps);
}

// ---- stage 3: y = w * (x - mean) * rstd + b ----
__global__ void bn_norm_k(const float* __restrict__ x,
                          const float* __restrict__ w,
                          const float* __restrict__ b,
                          const float* __restrict__ mean,
                          const float* __restrict__ rstd,
                          float* __restrict__ y, int Nb, int C, int HW) {
  long n = (long)Nb * C * HW;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int c = (i / HW) % C;
    y[i] = w[c] * (x[i] - mean[c]) * rstd[c] + b[c];
  }
}

// ---- bwd stage 1: per-channel sum(dy), sum(dy * xhat) ----
__global__ void bn_bwd_stats_k(const float* __restrict__ x,
                               const float* __restrict__ dy,
                               const float* __restrict__ mean,
                               const float* __restrict__ rstd,
                               float* __restrict__ out,  // [2][C]: db, dwdot
                               int Nb, int C, int HW) {
  int c = blockIdx.x;
  __shared__ float sh[2][kBlock];
  float s_dy = 0.f, s_dyx = 0.f;
  float m = mean[c], rs = rstd[c];
  for (long i = threadIdx.x; i < (long)Nb * HW; i += blockDim.x) {
    long nb = i / HW, px = i % HW;
    long idx = (nb * C + c) * (long)HW + px;
    float g = dy[idx];
    s_dy += g;
    s_dyx += g * (x[idx] - m) * rs;
  }
  sh[0][threadIdx.x] = s_dy;
  sh[1][threadIdx.x] = s_dyx;
  __syncthreads();
  for (int off = kBlock / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      sh[0][threadIdx.x] += sh[0][threadIdx.x + off];
      sh[1][threadIdx.x] += sh[1][threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    out[c] = sh[0][0];       // db
    out[C + c] = sh[1][0];   // sum(dy * xhat) = dw
  }
}

// ---- bwd stage 2: dx = (w*rstd/Nc)*(Nc*dy - db - xhat*dwdot) ----
__global__ void bn_bwd_dx_k(const float* __restrict__ x,
                            const float* __restrict__ dy,
                            const float* __restrict__ w,
                            const float* __restrict__ mean,
                            const float* __restrict__ rstd,
                            const float* __restrict__ stats, float*
                            __restrict__ dx, int Nb, int C, int HW,
                            int training) {
  long n = (long)Nb * C * HW;
  long stride = (long)gridDim.x * blockDim.x;
  float inv = 1.f / ((float)Nb * HW);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int c = (i / HW) % C;
    float g = dy[i];
    if (training) {
      float xhat = (x[i] - mean[c]) * rstd[c];
      dx[i] = w[c] * rstd[c] *
              (g - inv * (stats[c] + xhat * stats[C + c]));
    } else {
      dx[i] = w[c] * rstd[c] * g;
    }
  }
}

extern "C" {
void launch_bn_fwd(const float* x, const float* w, const float* b,
                   float* running_mean, float* running_var, float* save_mean,
                   float* save_rstd, float* y, float* scratch2C, int Nb,
                   int C, int HW, float momentum, float eps, int training,
                   void* s) {
  hipStream_t st = (hipStream_t)s;
  if (training) {
    bn_stats_k<<<C, kBlock, 0, st>>>(x, scratch2C, Nb, C, HW);
    bn_finalize_k<<<(C + 255) / 256, 256, 0, st>>>(
        scratch2C, save_mean, save_rstd, running_mean, running_var, C,
        (long)Nb * HW, momentum, eps);
  } else {
    bn_eval_stats_k<<<(C + 255) / 256, 256, 0, st>>>(
        running_mean, running_var, save_mean, save_rstd, C, eps);
  }
  bn_norm_k<<<grid_for((long)Nb * C * HW), kBlock, 0, st>>>(
      x, w, b, save_mean, save_rstd, y, Nb, C, HW);
}

void launch_bn_bwd(const float* x, const float* dy, const float* w,
                   const float* save_mean, const float* save_rstd,
                   float* stats2C, float* dx, float* dw, float* db, int Nb,
                   int C, int HW, int training, void* s) {
  hipStream_t st = (hipStream_t)s;
  bn_bwd_stats_k<<<C, kBlock, 0, st>>>(x, dy, save_mean, save_rstd, stats2C,
                                       Nb, C, HW);
  // db = stats[0:C], dw = stats[C:2C] — copied out by the binding
  bn_bwd_dx_k<<<grid_for((long)Nb * C * HW), kBlock, 0, st>>>(
      x, dy, w, save_mean, save_rstd, stats2C, dx, Nb, C, HW, training);
  HIP_CHECK(hipMemcpyAsync(db, stats2C, C * sizeof(float),
                           hipMemcpyDeviceToDevice, st));
  HIP_CHECK(hipMemcpyAsync(dw, stats2C + C, C * sizeof(float),
                           hipMemcpyDeviceToDevice, st));
}
}
