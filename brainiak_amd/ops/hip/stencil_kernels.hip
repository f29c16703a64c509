// Direct 3-D stencil aggregation for searchlight ball/cube kernels.
//
// torch/MIOpen lowers the [K,K,K] single-channel conv3d that
// aggregates per-voxel statistics over a searchlight ball to
// im2col + GEMM (~1/3 of the searchlight benchmark's device time);
// a K<=9 stencil wants a plain LDS-tiled sweep instead.
//
// x [B, X, Y, Z] fp32, w [K, K, K] fp32 (K = 2r+1), valid conv:
// out[b, i, j, k] = sum_ijk w * x[b, i.., j.., k..]
// Block: 8x8x8 output tile, (8+2r)^3 halo tile in LDS.

#include <hip/hip_runtime.h>

typedef long long ll;

#define BT 8          // output tile edge
#define MAXR 4        // radius <= 4 (K <= 9)

__global__ __launch_bounds__(512) void k_stencil3d(
    const float* __restrict__ x, const float* __restrict__ w,
    float* __restrict__ out, ll B, int X, int Y, int Z, int r) {
    const int K = 2 * r + 1;
    const int OX = X - 2 * r, OY = Y - 2 * r, OZ = Z - 2 * r;
    const int tx = (OX + BT - 1) / BT;
    const int ty = (OY + BT - 1) / BT;
    const int tz = (OZ + BT - 1) / BT;
    ll b = blockIdx.x;
    const int kb = (int)(b % tz); b /= tz;
    const int jb = (int)(b % ty); b /= ty;
    const int ib = (int)(b % tx); b /= tx;
    if (b >= B) return;
    const ll batch = b;

    const int T = BT + 2 * MAXR;           // 16
    __shared__ float tile[T][T][T];
    const int tid = threadIdx.x;
    const int span = BT + 2 * r;           // live halo extent

    // cooperative halo load (zero outside the volume)
    const float* xb = x + batch * (ll)X * Y * Z;
    for (int idx = tid; idx < span * span * span; idx += 512) {
        int lk = idx % span;
        int lj = (idx / span) % span;
        int li = idx / (span * span);
        int gi = ib * BT + li, gj = jb * BT + lj, gk = kb * BT + lk;
        float v = 0.0f;
        if (gi < X && gj < Y && gk < Z)
            v = xb[((ll)gi * Y + gj) * Z + gk];
        tile[li][lj][lk] = v;
    }
    __syncthreads();

    const int ok = tid % BT;
    const int oj = (tid / BT) % BT;
    const int oi = tid / (BT * BT);        // 512 = 8*8*8 exact
    const int go_i = ib * BT + oi, go_j = jb * BT + oj,
              go_k = kb * BT + ok;
    if (go_i >= OX || go_j >= OY || go_k >= OZ) return;

    float acc = 0.0f;
    for (int a = 0; a < K; ++a)
        for (int c = 0; c < K; ++c)
            for (int d = 0; d < K; ++d) {
                float wv = w[(a * K + c) * K + d];
                if (wv != 0.0f)
                    acc = fmaf(wv, tile[oi + a][oj + c][ok + d], acc);
            }
    out[((batch * OX + go_i) * (ll)OY + go_j) * OZ + go_k] = acc;
}

extern "C" void launch_stencil3d(const float* x, const float* w,
                                 float* out, ll B, int X, int Y, int Z,
                                 int r, hipStream_t stream) {
    int tx = (X - 2 * r + BT - 1) / BT;
    int ty = (Y - 2 * r + BT - 1) / BT;
    int tz = (Z - 2 * r + BT - 1) / BT;
    ll grid = B * (ll)tx * ty * tz;
    hipLaunchKernelGGL(k_stencil3d, dim3((unsigned)grid), dim3(512), 0,
                       stream, x, w, out, B, X, Y, Z, r);
}
