// Fused flat-buffer optimizer kernels.
// The reference runs clip_grad_norm_ + SGD.step + an optional full
// pack/norm/unpack PGD projection per BATCH (agent.py:50-60).  On flat
// buffers that whole sequence is two reductions + two updates; reductions
// are two-stage (block partials -> single-block finalize) for run-to-run
// determinism (no float atomics).
#include "common.h"

constexpr int kNPart = 1024;  // fixed partial count -> deterministic sum

// ---- stage 1: partial sum of squares (of g, or of (p - t0)) ----
__global__ void sq_partials_k(const float* __restrict__ a,
                              const float* __restrict__ sub,  // nullable
                              float* __restrict__ partials, long n) {
  __shared__ float sh[kBlock];
  float acc = 0.f;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float v = sub ? (a[i] - sub[i]) : a[i];
    acc += v * v;
  }
  sh[threadIdx.x] = acc;
  __syncthreads();
  for (int off = kBlock / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sh[threadIdx.x] += sh[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) partials[blockIdx.x] = sh[0];
}

// ---- stage 2: finalize the scalar the update kernels consume ----
// mode 0 (sgd clip):  out = min(1, max_norm / (||g|| + 1e-6))
// mode 1 (pgd):       out = 1 / max(1, ||u|| / clip)
__global__ void finalize_scale_k(const float* __restrict__ partials,
                                 float* __restrict__ out, int nb, float c,
                                 int mode) {
  __shared__ float sh[kNPart];
  float acc = (threadIdx.x < nb) ? partials[threadIdx.x] : 0.f;
  sh[threadIdx.x] = acc;
  __syncthreads();
  for (int off = kNPart / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sh[threadIdx.x] += sh[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    float norm = sqrtf(sh[0]);
    if (mode == 0) {
      float coef = c / (norm + 1e-6f);
      out[0] = coef < 1.f ? coef : 1.f;
    } else {
      float denom = norm / c;
      out[0] = denom > 1.f ? 1.f / denom : 1.f;
    }
  }
}

// ---- v = mu*v + g*scale ; p -= lr*v ----
__global__ void sgd_update_k(float* __restrict__ p, const float* __restrict__ g,
                             float* __restrict__ v,
                             const float* __restrict__ scale, float lr,
                             float mu, long n) {
  float s = scale[0];
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = n / 4;
  long i4 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long i = i4; i < n4; i += stride) {
    float4 gv = ((const float4*)g)[i];
    float4 vv = ((float4*)v)[i];
    float4 pv = ((float4*)p)[i];
    vv.x = mu * vv.x + gv.x * s; pv.x -= lr * vv.x;
    vv.y = mu * vv.y + gv.y * s; pv.y -= lr * vv.y;
    vv.z = mu * vv.z + gv.z * s; pv.z -= lr * vv.z;
    vv.w = mu * vv.w + gv.w * s; pv.w -= lr * vv.w;
    ((float4*)v)[i] = vv;
    ((float4*)p)[i] = pv;
  }
  for (long i = n4 * 4 + i4; i < n; i += stride) {
    float vv = mu * v[i] + g[i] * s;
    v[i] = vv;
    p[i] -= lr * vv;
  }
}

// ---- p = t0 + (p - t0) * scale  (PGD projection apply) ----
__global__ void pgd_apply_k(float* __restrict__ p,
                            const float* __restrict__ t0,
                            const float* __restrict__ scale, long n) {
  float s = scale[0];
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    p[i] = t0[i] + (p[i] - t0[i]) * s;
}

// ---- out = double(p) - theta0_64 ----
__global__ void delta64_k(const float* __restrict__ p,
                          const double* __restrict__ t0,
                          double* __restrict__ out, long n) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (double)p[i] - t0[i];
}

extern "C" {
void launch_clipped_sgd(float* p, const float* g, float* v, float* scratch,
                        float lr, float mu, float max_norm, long n, void* s) {
  // scratch: kNPart partials + 1 scale
  hipStream_t st = (hipStream_t)s;
  sq_partials_k<<<kNPart, kBlock, 0, st>>>(g, nullptr, scratch, n);
  finalize_scale_k<<<1, kNPart, 0, st>>>(scratch, scratch + kNPart, kNPart,
                                         max_norm, 0);
  sgd_update_k<<<grid_for(n / 4 + 1), kBlock, 0, st>>>(p, g, v,
                                                       scratch + kNPart, lr,
                                                       mu, n);
}
void launch_pgd_project(float* p, const float* t0, float* scratch, float clip,
                        long n, void* s) {
  hipStream_t st = (hipStream_t)s;
  sq_partials_k<<<kNPart, kBlock, 0, st>>>(p, t0, scratch, n);
  finalize_scale_k<<<1, kNPart, 0, st>>>(scratch, scratch + kNPart, kNPart,
                                         clip, 1);
  pgd_apply_k<<<grid_for(n), kBlock, 0, st>>>(p, t0, scratch + kNPart, n);
}
void launch_delta64(const float* p, const double* t0, double* out, long n,
                    void* s) {
  delta64_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(p, t0, out, n);
}
}

// ---- multi-segment gather: scattered autograd grad tensors -> the flat
// grad buffer in ONE launch (replaces zero_grad + one accumulate-add per
// parameter; pointers ride in kernargs so the kernel is hipGraph-stable).
struct GatherSegs {
  const float* src[16];
  long off[16];
  long len[16];
  int count;
};

__global__ void gather_grads_k(GatherSegs segs, float* __restrict__ flat) {
  long total = 0;
#pragma unroll
  for (int i = 0; i < 16; ++i)
    if (i < segs.count) total += segs.len[i];
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += stride) {
    long rem = t;
    int seg = 0;
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      if (i < segs.count && rem >= segs.len[i]) {
        rem -= segs.len[i];
        seg = i + 1;
      }
    }
    flat[segs.off[seg] + rem] = segs.src[seg][rem];
  }
}

extern "C" void launch_gather_grads(const float* const* srcs,
                                    const long* offs, const long* lens,
                                    int count, float* flat, void* s) {
  GatherSegs g;
  g.count = count;
  long total = 0;
  for (int i = 0; i < count && i < 16; ++i) {
    g.src[i] = srcs[i];
    g.off[i] = offs[i];
    g.len[i] = lens[i];
    total += lens[i];
  }
  gather_grads_k<<<grid_for(total), kBlock, 0, (hipStream_t)s>>>(g, flat);
}
