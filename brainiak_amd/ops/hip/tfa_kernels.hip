// TFA kernels (MI355X equivalents of the reference's OpenMP extension,
// ref src/brainiak/factoranalysis/tfa_extension.cpp:28-239):
//
//  k_tfa_factor: RBF factor matrix F[v,k] = exp(-||coord_v - center_k||^2
//                / width_k).  The reference exploits unique-coordinate
//                tables; on gfx950 the direct form is trivially
//                memory/VALU-bound with centers staged in LDS, and keeps
//                arbitrary (non-grid) coordinates working.
//  k_tfa_recon:  residual R[v,t] = scale * (X[v,t] - sum_k F[v,k] W[k,t])
//                flattened, the other hot op of TFA's least-squares loop.

#include <hip/hip_runtime.h>

typedef long long ll;

#define TFA_MAXK 128

__global__ __launch_bounds__(256) void k_tfa_factor(
    const float* __restrict__ centers,  // [K, 3]
    const float* __restrict__ widths,   // [K]
    const float* __restrict__ coords,   // [V, 3]
    float* __restrict__ F,              // [V, K]
    ll V, int K) {
    __shared__ float cx[TFA_MAXK], cy[TFA_MAXK], cz[TFA_MAXK],
        winv[TFA_MAXK];
    for (int k = threadIdx.x; k < K; k += 256) {
        cx[k] = centers[3 * k + 0];
        cy[k] = centers[3 * k + 1];
        cz[k] = centers[3 * k + 2];
        winv[k] = 1.0f / widths[k];
    }
    __syncthreads();
    const ll v = (ll)blockIdx.x * 256 + threadIdx.x;
    if (v >= V) return;
    const float x = coords[3 * v + 0];
    const float y = coords[3 * v + 1];
    const float z = coords[3 * v + 2];
    float* out = F + v * K;
    for (int k = 0; k < K; ++k) {
        float dx = x - cx[k], dy = y - cy[k], dz = z - cz[k];
        out[k] = __expf(-(dx * dx + dy * dy + dz * dz) * winv[k]);
    }
}

__global__ __launch_bounds__(256) void k_tfa_recon(
    const float* __restrict__ X,   // [V, T]
    const float* __restrict__ W,   // [K, T]
    const float* __restrict__ F,   // [V, K]
    float* __restrict__ R,         // [V * T]
    ll V, ll T, int K, float scale) {
    const ll idx = (ll)blockIdx.x * 256 + threadIdx.x;
    if (idx >= V * T) return;
    const ll v = idx / T;
    const ll t = idx % T;
    const float* f = F + v * K;
    float acc = 0.0f;
    for (int k = 0; k < K; ++k)
        acc = fmaf(f[k], W[(ll)k * T + t], acc);
    R[idx] = scale * (X[idx] - acc);
}

extern "C" void launch_tfa_factor(const float* centers, const float* widths,
                                  const float* coords, float* F, ll V,
                                  int K, hipStream_t stream) {
    ll grid = (V + 255) / 256;
    hipLaunchKernelGGL(k_tfa_factor, dim3((unsigned)grid), dim3(256), 0,
                       stream, centers, widths, coords, F, V, K);
}

extern "C" void launch_tfa_recon(const float* X, const float* W,
                                 const float* F, float* R, ll V, ll T,
                                 int K, float scale, hipStream_t stream) {
    ll grid = (V * T + 255) / 256;
    hipLaunchKernelGGL(k_tfa_recon, dim3((unsigned)grid), dim3(256), 0,
                       stream, X, W, F, R, V, T, K, scale);
}
