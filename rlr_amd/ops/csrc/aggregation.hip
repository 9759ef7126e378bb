// Server aggregation kernels (reference
// aggregation.py:19-75).  All operate on the stacked (K, n_params) fp64
// update matrix in sampled order; per-coordinate K-loops run in that fixed
// order, so results are bitwise identical for any world size and any run.
//
// The headline path (avg + RLR + apply, reference aggregation.py:21-40) is
// ONE fused kernel: a single pass over the K x n fp64 matrix produces the
// sign-vote LR, the weighted average, optional philox gaussian noise and
// the fp32 parameter write — the matrix is read once instead of three
// times (HBM-bound: K x n x 8 B per pass).
#include "common.h"

// ---------------------------------------------------- fused avg+RLR+apply
// mode_rlr: 0 = constant server_lr, 1 = RLR vote
// noise_std <= 0 disables noise.
__global__ void fused_avg_rlr_apply_k(
    const double* __restrict__ U, const double* __restrict__ w, int K, long n,
    double inv_wsum, int mode_rlr, double thresh, double slr,
    float* __restrict__ params, double* __restrict__ lr_out,
    double noise_std, uint64_t seed, uint64_t offset) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double acc = 0.0;
    double votes = 0.0;
    for (int k = 0; k < K; ++k) {
      double u = U[(long)k * n + i];
      acc += w[k] * u;
      votes += (u > 0.0) ? 1.0 : (u < 0.0 ? -1.0 : 0.0);
    }
    double avg = acc * inv_wsum;
    double lr = slr;
    if (mode_rlr) lr = (fabs(votes) >= thresh) ? slr : -slr;
    if (lr_out) lr_out[i] = lr;
    if (noise_std > 0.0) {
      // Box-Muller from philox (deterministic per (seed, round, i))
      Philox4 r = philox4(seed, offset, (uint32_t)i);
      double u1 = (double)u32_to_uniform(r.x);
      double u2 = (double)u32_to_uniform(r.y);
      avg += noise_std * sqrt(-2.0 * log(u1)) * cos(6.283185307179586 * u2);
    }
    params[i] = (float)((double)params[i] + lr * avg);
  }
}

// ------------------------------------------------------ standalone pieces

__global__ void rlr_vote_k(const double* __restrict__ U, int K, long n,
                           double thresh, double slr,
                           double* __restrict__ lr) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double votes = 0.0;
    for (int k = 0; k < K; ++k) {
      double u = U[(long)k * n + i];
      votes += (u > 0.0) ? 1.0 : (u < 0.0 ? -1.0 : 0.0);
    }
    lr[i] = (fabs(votes) >= thresh) ? slr : -slr;
  }
}

__global__ void agg_avg_k(const double* __restrict__ U,
                          const double* __restrict__ w, int K, long n,
                          double inv_wsum, double* __restrict__ out) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double acc = 0.0;
    for (int k = 0; k < K; ++k) acc += w[k] * U[(long)k * n + i];
    out[i] = acc * inv_wsum;
  }
}

__global__ void agg_sign_k(const double* __restrict__ U, int K, long n,
                           double* __restrict__ out) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double votes = 0.0;
    for (int k = 0; k < K; ++k) {
      double u = U[(long)k * n + i];
      votes += (u > 0.0) ? 1.0 : (u < 0.0 ? -1.0 : 0.0);
    }
    out[i] = (votes > 0.0) ? 1.0 : (votes < 0.0 ? -1.0 : 0.0);
  }
}

// Coordinate median with torch.median semantics: the LOWER of the two
// middle values for even K = the ((K-1)/2)-th smallest.  K <= 64.
// Selection by repeated min-extraction over registers via a "count smaller,
// then pick" scheme: O(K^2) compares in registers, no scratch arrays
// (runtime-indexed local arrays spill — cdna_hip_programming.md §5.4 #20).
__global__ void agg_comed_k(const double* __restrict__ U, int K, long n,
                            double* __restrict__ out) {
  long stride = (long)gridDim.x * blockDim.x;
  int target = (K - 1) / 2;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double med = 0.0;
    for (int k = 0; k < K; ++k) {
      double v = U[(long)k * n + i];
      int smaller = 0, equal_before = 0;
      for (int j = 0; j < K; ++j) {
        double u = U[(long)j * n + i];
        if (u < v || (u == v && j < k)) {
          if (u < v) smaller++;
          else equal_before++;
        }
      }
      // rank of element k in a stable sort
      if (smaller + equal_before == target) med = v;
    }
    out[i] = med;
  }
}

__global__ void apply_update_k(float* __restrict__ params,
                               const double* __restrict__ agg,
                               const double* __restrict__ lr,  // nullable
                               double slr, long n) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double l = lr ? lr[i] : slr;
    params[i] = (float)((double)params[i] + l * agg[i]);
  }
}

__global__ void add_noise_k(double* __restrict__ agg, double std, long n,
                            uint64_t seed, uint64_t offset) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    Philox4 r = philox4(seed, offset, (uint32_t)i);
    double u1 = (double)u32_to_uniform(r.x);
    double u2 = (double)u32_to_uniform(r.y);
    agg[i] += std * sqrt(-2.0 * log(u1)) * cos(6.283185307179586 * u2);
  }
}

extern "C" {
void launch_fused_avg_rlr_apply(const double* U, const double* w, int K,
                                long n, double inv_wsum, int mode_rlr,
                                double thresh, double slr, float* params,
                                double* lr_out, double noise_std,
                                uint64_t seed, uint64_t offset, void* s) {
  fused_avg_rlr_apply_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(
      U, w, K, n, inv_wsum, mode_rlr, thresh, slr, params, lr_out, noise_std,
      seed, offset);
}
void launch_rlr_vote(const double* U, int K, long n, double thresh,
                     double slr, double* lr, void* s) {
  rlr_vote_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(U, K, n, thresh, slr,
                                                         lr);
}
void launch_agg_avg(const double* U, const double* w, int K, long n,
                    double inv_wsum, double* out, void* s) {
  agg_avg_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(U, w, K, n, inv_wsum,
                                                        out);
}
void launch_agg_sign(const double* U, int K, long n, double* out, void* s) {
  agg_sign_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(U, K, n, out);
}
void launch_agg_comed(const double* U, int K, long n, double* out, void* s) {
  agg_comed_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(U, K, n, out);
}
void launch_apply_update(float* params, const double* agg, const double* lr,
                         double slr, long n, void* s) {
  apply_update_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(params, agg, lr,
                                                             slr, n);
}
void launch_add_noise(double* agg, double std, long n, uint64_t seed,
                      uint64_t offset, void* s) {
  add_noise_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(agg, std, n, seed,
                                                          offset);
}
}
