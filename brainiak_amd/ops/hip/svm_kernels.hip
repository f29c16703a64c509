// Batched precomputed-kernel C-SVC cross-validation, fully on-device.
//
// The reference scores each voxel's [E,E] kernel with sklearn SVC in a CPU
// process pool (ref src/brainiak/fcma/voxelselector.py:423-465).  Here ONE
// kernel launch solves every (voxel, fold) dual QP: one workgroup (64
// threads = 1 wave) per problem, Q staged in LDS, SMO with
// maximal-violating-pair selection (LIBSVM WSS1), then the fold's test
// accuracy computed in the same block.  n_train <= 64 (one dual variable
// per lane).
//
// Output: correct-prediction counts [C, F] int32; host divides by test
// counts and averages folds.

#include <hip/hip_runtime.h>

typedef long long ll;

#define SVM_MAXN 64

__global__ __launch_bounds__(64) void k_svm_cv(
    const float* __restrict__ kernels,  // [C, E, E]
    const float* __restrict__ y,        // [E] in {-1, +1}
    const int* __restrict__ train_idx,  // [F, E] (first n_train valid)
    const int* __restrict__ test_idx,   // [F, E] (first n_test valid)
    const int* __restrict__ n_train,    // [F]
    const int* __restrict__ n_test,     // [F]
    int* __restrict__ correct,          // [C, F]
    ll C, int E, int F, float Creg, float tol, int max_iter) {
    const ll blk = blockIdx.x;
    const ll c = blk / F;
    const int f = (int)(blk % F);
    if (c >= C) return;
    const int lane = threadIdx.x;
    const int n = n_train[f];
    const int m = n_test[f];

    __shared__ float Q[SVM_MAXN][SVM_MAXN + 1];
    __shared__ float ys[SVM_MAXN];
    __shared__ float alpha_s[SVM_MAXN];
    __shared__ int pair[2];
    __shared__ float delta[2];

    const float* Kc = kernels + c * (ll)E * E;
    const int* tr = train_idx + (ll)f * E;

    // stage labels + Q = y_i y_j K[tr_i][tr_j]
    float yl = 0.0f;
    if (lane < n) {
        yl = y[tr[lane]];
        ys[lane] = yl;
    }
    __syncthreads();
    if (lane < n) {
        const float* krow = Kc + (ll)tr[lane] * E;
        for (int j = 0; j < n; ++j)
            Q[lane][j] = yl * ys[j] * krow[tr[j]];
    }
    __syncthreads();

    // SMO: one dual variable per lane
    float alpha = 0.0f;
    float grad = -1.0f;           // (Q alpha)_i - 1
    const bool valid = lane < n;

    float b = 0.0f;
    for (int iter = 0; iter < max_iter; ++iter) {
        float score = -yl * grad;
        bool in_up = valid && ((yl > 0.f && alpha < Creg - 1e-12f) ||
                               (yl < 0.f && alpha > 1e-12f));
        bool in_low = valid && ((yl > 0.f && alpha > 1e-12f) ||
                                (yl < 0.f && alpha < Creg - 1e-12f));
        // wave argmax over I_up / argmin over I_low
        float up = in_up ? score : -3.0e38f;
        float lo = in_low ? score : 3.0e38f;
        int up_i = lane, lo_i = lane;
        #pragma unroll
        for (int d = 32; d > 0; d >>= 1) {
            float u2 = __shfl_xor(up, d);
            int ui2 = __shfl_xor(up_i, d);
            if (u2 > up) { up = u2; up_i = ui2; }
            float l2 = __shfl_xor(lo, d);
            int li2 = __shfl_xor(lo_i, d);
            if (l2 < lo) { lo = l2; lo_i = li2; }
        }
        if (up - lo < tol) { b = 0.5f * (up + lo); break; }
        b = 0.5f * (up + lo);

        // publish lane-private alpha so lane 0 can read the pair's values
        if (valid) alpha_s[lane] = alpha;
        __syncthreads();
        if (lane == 0) {
            int i = up_i, j = lo_i;
            float yi = ys[i], yj = ys[j];
            float eta = Q[i][i] + Q[j][j] - 2.0f * Q[i][j] * yi * yj;
            eta = fmaxf(eta, 1e-12f);
            float t = (up - lo) / eta;
            float ai = alpha_s[i], aj = alpha_s[j];
            float tmax_i = (yi > 0.f) ? (Creg - ai) : ai;
            float tmax_j = (yj > 0.f) ? aj : (Creg - aj);
            t = fminf(t, fminf(tmax_i, tmax_j));
            t = fmaxf(t, 0.0f);
            pair[0] = i; pair[1] = j;
            delta[0] = yi * t;      // d alpha_i
            delta[1] = -yj * t;     // d alpha_j
        }
        __syncthreads();
        const int i = pair[0], j = pair[1];
        const float dai = delta[0], daj = delta[1];
        if (lane == i) alpha += dai;
        if (lane == j) alpha += daj;
        if (valid) grad += Q[i][lane] * dai + Q[j][lane] * daj;
        __syncthreads();
    }

    // publish final alpha, predict test samples
    if (valid) alpha_s[lane] = alpha * yl;   // coef_i = alpha_i * y_i
    __syncthreads();
    const int* te = test_idx + (ll)f * E;
    int correct_local = 0;
    if (lane < m) {
        const float* krow = Kc + (ll)te[lane] * E;
        float dec = b;
        for (int i = 0; i < n; ++i)
            dec = fmaf(alpha_s[i], krow[tr[i]], dec);
        float yt = y[te[lane]];
        correct_local = ((dec > 0.f) == (yt > 0.f)) ? 1 : 0;
    }
    #pragma unroll
    for (int d = 32; d > 0; d >>= 1)
        correct_local += __shfl_xor(correct_local, d);
    if (lane == 0) correct[c * F + f] = correct_local;
}

extern "C" void launch_svm_cv(const float* kernels, const float* y,
                              const int* train_idx, const int* test_idx,
                              const int* n_train, const int* n_test,
                              int* correct, ll C, int E, int F, float Creg,
                              float tol, int max_iter, hipStream_t stream) {
    hipLaunchKernelGGL(k_svm_cv, dim3((unsigned)(C * F)), dim3(64), 0,
                       stream, kernels, y, train_idx, test_idx, n_train,
                       n_test, correct, C, E, F, Creg, tol, max_iter);
}
