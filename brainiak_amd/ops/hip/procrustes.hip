// Batched K x K symmetric Jacobi eigensolver for SRM's Procrustes update
// (MI355X replacement for the per-subject LAPACK SVD the reference runs on
// CPU, ref src/brainiak/funcalign/srm.py:595-606).
//
// The Procrustes factor W = U V^T of A [V, K] equals the polar factor
// A (A^T A)^{-1/2}; A^T A is K x K (K = features ~ 50), so the device-side
// work is V-independent: the host computes G = A^T A (one rocBLAS gemm on
// MFMA), this kernel eigensolves every G in the batch in LDS, and the host
// assembles A . (Vec diag(rsqrt(lambda)) Vec^T) with a second gemm.
//
// One block = one wave (64 threads) per matrix; classic cyclic Jacobi with
// wave-parallel row/column rotation updates; K <= 64.

#include <hip/hip_runtime.h>

typedef long long ll;

#define JAC_MAXK 64
#define JAC_SWEEPS 10
#define JAC_EPS 1e-12f

__global__ __launch_bounds__(64) void k_jacobi_eigh(
    const float* __restrict__ Gin,   // [B, K, K]
    float* __restrict__ evecs,       // [B, K, K]  (columns = eigenvectors)
    float* __restrict__ evals,       // [B, K]
    ll B, int K) {
    const ll b = blockIdx.x;
    if (b >= B) return;
    const int t = threadIdx.x;

    __shared__ float Gs[JAC_MAXK][JAC_MAXK + 1];
    __shared__ float Vs[JAC_MAXK][JAC_MAXK + 1];
    __shared__ float rot[4];           // c, s, off-diag magnitude, unused

    const float* G0 = Gin + b * (ll)K * K;
    for (int idx = t; idx < K * K; idx += 64) {
        int i = idx / K, j = idx % K;
        Gs[i][j] = G0[idx];
        Vs[i][j] = (i == j) ? 1.0f : 0.0f;
    }
    __syncthreads();

    for (int sweep = 0; sweep < JAC_SWEEPS; ++sweep) {
        // convergence check: max |off-diagonal|
        float local_off = 0.0f;
        for (int idx = t; idx < K * K; idx += 64) {
            int i = idx / K, j = idx % K;
            if (i < j) local_off = fmaxf(local_off, fabsf(Gs[i][j]));
        }
        #pragma unroll
        for (int d = 32; d > 0; d >>= 1)
            local_off = fmaxf(local_off, __shfl_xor(local_off, d));
        if (t == 0) rot[2] = local_off;
        __syncthreads();
        if (rot[2] < 1e-8f) break;

        for (int p = 0; p < K - 1; ++p) {
            for (int q = p + 1; q < K; ++q) {
                if (t == 0) {
                    float apq = Gs[p][q];
                    float app = Gs[p][p], aqq = Gs[q][q];
                    float c = 1.0f, s = 0.0f;
                    if (fabsf(apq) > JAC_EPS * sqrtf(fabsf(app * aqq)
                                                     + 1e-30f)) {
                        float tau = (aqq - app) / (2.0f * apq);
                        float tt = (tau >= 0.f)
                            ? 1.0f / (tau + sqrtf(1.0f + tau * tau))
                            : 1.0f / (tau - sqrtf(1.0f + tau * tau));
                        c = rsqrtf(1.0f + tt * tt);
                        s = tt * c;
                    }
                    rot[0] = c; rot[1] = s;
                }
                __syncthreads();
                float c = rot[0], s = rot[1];
                if (s != 0.0f) {
                    // rows p,q of G (columns follow by symmetry)
                    if (t < K) {
                        float gp = Gs[p][t], gq = Gs[q][t];
                        Gs[p][t] = c * gp - s * gq;
                        Gs[q][t] = s * gp + c * gq;
                    }
                    __syncthreads();
                    if (t < K) {
                        float gp = Gs[t][p], gq = Gs[t][q];
                        Gs[t][p] = c * gp - s * gq;
                        Gs[t][q] = s * gp + c * gq;
                        float vp = Vs[t][p], vq = Vs[t][q];
                        Vs[t][p] = c * vp - s * vq;
                        Vs[t][q] = s * vp + c * vq;
                    }
                }
                __syncthreads();
            }
        }
    }

    float* V0 = evecs + b * (ll)K * K;
    for (int idx = t; idx < K * K; idx += 64)
        V0[idx] = Vs[idx / K][idx % K];
    if (t < K) evals[b * K + t] = Gs[t][t];
}

extern "C" void launch_jacobi_eigh(const float* G, float* evecs,
                                   float* evals, ll B, int K,
                                   hipStream_t stream) {
    hipLaunchKernelGGL(k_jacobi_eigh, dim3((unsigned)B), dim3(64), 0,
                       stream, G, evecs, evals, B, K);
}
