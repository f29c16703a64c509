// Fused ISFC accumulation: acc += atanh(clamp((M + M^T) / 2)) over a
// [V, V] correlation matrix, one pass.
//
// The torch expression behind isfc_distributed's Fisher-mean branch
// (ref brainiak/isc.py:211-373 semantics) materializes M^T, the
// clamp, the atanh and the add as separate [V, V] passes (~80 GB of
// HBM traffic per subject at V=50k); this kernel reads M once, loads
// the mirror tile through LDS for the transpose, and read-modify-
// writes acc — ~30 GB per subject.
//
// Tiles: 64x64, one block per upper-triangle tile pair (i <= j); the
// block also writes the mirrored tile.  Clamp matches torch:
// atanh(clamp(x, -1+1e-7, 1-1e-7)).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef long long ll;
typedef __hip_bfloat16 bf16_t;

#define IT 64     // tile edge
#define IPAD 1

__device__ __forceinline__ float atanh_clamped(float x) {
    const float lim = 1.0f - 1e-7f;
    x = fminf(fmaxf(x, -lim), lim);
    return 0.5f * (__logf(1.0f + x) - __logf(1.0f - x));
}

// B stacked matrices amortize the acc read-modify-write: per tile the
// acc loads/stores happen ONCE while B subjects' M tiles stream
// through (per-subject HBM drops from ~30 GB to ~10 GB + 20/B GB).
template <typename T>
__global__ __launch_bounds__(256) void k_isfc_accum(
    float* __restrict__ acc, const T* __restrict__ M, ll V, ll B) {
    const ll tiles = (V + IT - 1) / IT;
    // upper-triangle tile index -> (ti, tj), ti <= tj
    ll b = blockIdx.x;
    ll ti = (ll)((-1.0 + sqrt(1.0 + 8.0 * (double)b)) / 2.0);
    // fix rounding
    while ((ti + 1) * (ti + 2) / 2 <= b) ++ti;
    while (ti * (ti + 1) / 2 > b) --ti;
    const ll tj = b - ti * (ti + 1) / 2;      // tj <= ti
    const ll I0 = tj * IT;                    // row block (<= col block)
    const ll J0 = ti * IT;
    if (ti >= tiles) return;

    // ONE fp32 tile, two lives: the mirror tile is pre-read into
    // registers (16 values/thread), then the same LDS backs the z
    // tile for the transpose accumulation.  Two fp32 tiles (33 KB)
    // capped the kernel at 4 blocks/CU = 16 waves and PMC showed
    // 85 % memory-wait — halving LDS doubles residency with NO
    // precision change (a bf16-tile variant hit 1e-4-level errors).
    __shared__ float mt[IT][IT + IPAD];
    const int tx = threadIdx.x & 63;
    const int ty = threadIdx.x >> 6;          // 4 rows per pass
    const bool diag = (I0 == J0);

    // per-thread accumulators over the subject batch (16 rows/thread)
    float zsum[IT / 4];
    float zsum_m[IT / 4];
    #pragma unroll
    for (int q = 0; q < IT / 4; ++q) { zsum[q] = 0.f; zsum_m[q] = 0.f; }

    for (ll bm = 0; bm < B; ++bm) {
        const T* Mb = M + bm * V * V;
        // stage the mirror tile (rows J0..), coalesced over tx
        #pragma unroll
        for (int r = ty; r < IT; r += 4) {
            ll row = J0 + r, col = I0 + tx;
            mt[r][tx] = (row < V && col < V)
                        ? (float)Mb[row * V + col] : 0.0f;
        }
        __syncthreads();
        // pre-read this thread's transposed mirror values so the
        // tile can be reused for z below
        float mirr[IT / 4];
        #pragma unroll
        for (int r = ty, q = 0; r < IT; r += 4, ++q)
            mirr[q] = mt[tx][r];
        __syncthreads();
        #pragma unroll
        for (int r = ty, q = 0; r < IT; r += 4, ++q) {
            ll row = I0 + r, col = J0 + tx;
            float z = 0.0f;
            if (row < V && col < V) {
                float a = (float)Mb[row * V + col];
                float sym = 0.5f * (a + mirr[q]);
                z = atanh_clamped(sym);
                zsum[q] += z;
            }
            mt[r][tx] = z;                    // second life: z tile
        }
        if (!diag) {
            __syncthreads();
            #pragma unroll
            for (int r = ty, q = 0; r < IT; r += 4, ++q)
                zsum_m[q] += mt[tx][r];
        }
        __syncthreads();   // tile reused next bm
    }

    #pragma unroll
    for (int r = ty, q = 0; r < IT; r += 4, ++q) {
        ll row = I0 + r, col = J0 + tx;
        if (row < V && col < V)
            acc[row * V + col] += zsum[q];
    }
    if (diag) return;
    // mirror tile written COALESCED (a direct acc[col*V+row] scatter
    // was 3.5x the HBM roofline)
    #pragma unroll
    for (int r = ty, q = 0; r < IT; r += 4, ++q) {
        ll row = J0 + r, col = I0 + tx;
        if (row < V && col < V)
            acc[row * V + col] += zsum_m[q];
    }
}

extern "C" void launch_isfc_accum(float* acc, const void* M, ll V,
                                  ll B, int m_is_bf16,
                                  hipStream_t stream) {
    ll tiles = (V + IT - 1) / IT;
    ll nblocks = tiles * (tiles + 1) / 2;
    if (m_is_bf16)
        hipLaunchKernelGGL(k_isfc_accum<bf16_t>, dim3((unsigned)nblocks),
                           dim3(256), 0, stream, acc, (const bf16_t*)M,
                           V, B);
    else
        hipLaunchKernelGGL(k_isfc_accum<float>, dim3((unsigned)nblocks),
                           dim3(256), 0, stream, acc, (const float*)M,
                           V, B);
}
