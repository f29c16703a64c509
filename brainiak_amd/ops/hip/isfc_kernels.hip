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

// ===========================================================================
// k_isfc_fused: the whole per-subject ISFC update WITHOUT materializing
// the [V, V] correlation matrix.  For every upper-triangle 64x64 tile
// pair it computes BOTH correlation tiles D1 = Zs[I]·Zm[J]^T and
// D2 = Zs[J]·Zm[I]^T with bf16 MFMA (K = TRs), symmetrizes
// (sym = (D1 + D2^T)/2), applies atanh in registers, accumulates the
// per-subject z sums in VGPRs across the whole subject batch, and
// read-modify-writes acc ONCE per tile at the end.  The M write
// (~10 GB/subject) and its re-read by the accumulation kernel both
// disappear from HBM.
//
// Numerics match the precision='bf16' GEMM path (bf16 operands, fp32
// accumulation); the exact-fp32 path keeps the GEMM + k_isfc_accum
// pipeline.
// ===========================================================================
typedef short bf16x8f __attribute__((ext_vector_type(8)));
typedef float f32x4f __attribute__((ext_vector_type(4)));

#define IF_FK 32
#define IF_FKP (IF_FK + 8)

__global__ __launch_bounds__(256) void k_isfc_fused(
    float* __restrict__ acc, const bf16_t* __restrict__ Zs,
    const bf16_t* __restrict__ Zm, ll V, ll T, ll B) {
    const ll tiles = (V + IT - 1) / IT;
    ll b = blockIdx.x;
    ll ti = (ll)((-1.0 + sqrt(1.0 + 8.0 * (double)b)) / 2.0);
    while ((ti + 1) * (ti + 2) / 2 <= b) ++ti;
    while (ti * (ti + 1) / 2 > b) --ti;
    const ll tj = b - ti * (ti + 1) / 2;
    const ll I0 = tj * IT;                    // row block
    const ll J0 = ti * IT;                    // col block
    if (ti >= tiles) return;
    const bool diag = (I0 == J0);

    // union LDS: 4 staging tiles (20.5 KB) / one fp32 64x68 tile
    __shared__ char smem[4 * IT * IF_FKP * 2 > IT * 68 * 4
                         ? 4 * IT * IF_FKP * 2 : IT * 68 * 4];
    bf16_t (*st)[IT][IF_FKP] = (bf16_t (*)[IT][IF_FKP])smem;
    float (*ft)[68] = (float (*)[68])smem;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;
    const int wr = (w >> 1) * 32;
    const int wc = (w & 1) * 32;
    const int frow = lane & 15;
    const int fk = 8 * (lane >> 4);
    const int drow = (lane >> 4) * 4;
    const int dcol = lane & 15;

    float zsum[16];
    #pragma unroll
    for (int q = 0; q < 16; ++q) zsum[q] = 0.0f;

    const int kts = (int)((T + IF_FK - 1) / IF_FK);
    for (ll bm = 0; bm < B; ++bm) {
        const bf16_t* zs = Zs + bm * V * T;
        const bf16_t* zm = Zm + bm * V * T;
        f32x4f a1[4], a2[4];
        #pragma unroll
        for (int q = 0; q < 4; ++q) {
            a1[q] = (f32x4f)0.f;
            a2[q] = (f32x4f)0.f;
        }
        for (int kt = 0; kt < kts; ++kt) {
            const ll k0 = (ll)kt * IF_FK;
            __syncthreads();                  // LDS free (prev reads)
            {   // stage 4 tiles: tile t rows from (I0|J0) of (Zs|Zm)
                const int tile = tid >> 6;
                const int row = tid & 63;
                // tiles: 0 = Zs[I], 1 = Zm[J], 2 = Zs[J], 3 = Zm[I]
                const bf16_t* src = (tile == 0 || tile == 2) ? zs : zm;
                const ll r = ((tile == 0 || tile == 3) ? I0 : J0)
                             + row;
                if (r < V && k0 + IF_FK <= T) {
                    const bf16x8f* p8 = (const bf16x8f*)
                        &src[r * T + k0];
                    #pragma unroll
                    for (int h = 0; h < IF_FK / 8; ++h)
                        *(bf16x8f*)&st[tile][row][8 * h] = p8[h];
                } else {
                    #pragma unroll
                    for (int c = 0; c < IF_FK; ++c) {
                        ll col = k0 + c;
                        st[tile][row][c] = (r < V && col < T)
                            ? src[r * T + col] : (bf16_t)0.0f;
                    }
                }
            }
            __syncthreads();
            bf16x8f fi0 = *(const bf16x8f*)&st[0][wr + frow][fk];
            bf16x8f fi1 = *(const bf16x8f*)&st[0][wr + 16 + frow][fk];
            bf16x8f fj0 = *(const bf16x8f*)&st[1][wc + frow][fk];
            bf16x8f fj1 = *(const bf16x8f*)&st[1][wc + 16 + frow][fk];
            a1[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                fi0, fj0, a1[0], 0, 0, 0);
            a1[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                fi0, fj1, a1[1], 0, 0, 0);
            a1[2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                fi1, fj0, a1[2], 0, 0, 0);
            a1[3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                fi1, fj1, a1[3], 0, 0, 0);
            if (!diag) {
                bf16x8f gi0 = *(const bf16x8f*)&st[2][wr + frow][fk];
                bf16x8f gi1 =
                    *(const bf16x8f*)&st[2][wr + 16 + frow][fk];
                bf16x8f gj0 = *(const bf16x8f*)&st[3][wc + frow][fk];
                bf16x8f gj1 =
                    *(const bf16x8f*)&st[3][wc + 16 + frow][fk];
                a2[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    gi0, gj0, a2[0], 0, 0, 0);
                a2[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    gi0, gj1, a2[1], 0, 0, 0);
                a2[2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    gi1, gj0, a2[2], 0, 0, 0);
                a2[3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    gi1, gj1, a2[3], 0, 0, 0);
            }
        }
        // epilogue: D2 (or D1 on the diagonal) through LDS for the
        // transpose, then sym+atanh accumulated in registers
        __syncthreads();
        const f32x4f* d2 = diag ? a1 : a2;
        #pragma unroll
        for (int q = 0; q < 4; ++q) {
            int mi = q >> 1, ni = q & 1;
            #pragma unroll
            for (int r = 0; r < 4; ++r)
                ft[wr + 16 * mi + drow + r][wc + 16 * ni + dcol] =
                    d2[q][r];
        }
        __syncthreads();
        #pragma unroll
        for (int q = 0; q < 4; ++q) {
            int mi = q >> 1, ni = q & 1;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                int gr = wr + 16 * mi + drow + r;
                int gc = wc + 16 * ni + dcol;
                float sym = 0.5f * (a1[q][r] + ft[gc][gr]);
                zsum[q * 4 + r] += atanh_clamped(sym);
            }
        }
    }

    // one acc read-modify-write per tile (plus the mirrored tile)
    __syncthreads();
    #pragma unroll
    for (int q = 0; q < 4; ++q) {
        int mi = q >> 1, ni = q & 1;
        #pragma unroll
        for (int r = 0; r < 4; ++r)
            ft[wr + 16 * mi + drow + r][wc + 16 * ni + dcol] =
                zsum[q * 4 + r];
    }
    __syncthreads();
    #pragma unroll
    for (int q = 0; q < 4; ++q) {
        int mi = q >> 1, ni = q & 1;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int gr = wr + 16 * mi + drow + r;
            int gc = wc + 16 * ni + dcol;
            ll grow = I0 + gr, gcol = J0 + gc;
            if (grow < V && gcol < V)
                acc[grow * V + gcol] += zsum[q * 4 + r];
            if (!diag) {
                ll mrow = J0 + gr, mcol = I0 + gc;
                if (mrow < V && mcol < V)
                    acc[mrow * V + mcol] += ft[gc][gr];
            }
        }
    }
}

extern "C" void launch_isfc_fused(float* acc, const void* Zs,
                                  const void* Zm, ll V, ll T, ll B,
                                  hipStream_t stream) {
    ll tiles = (V + IT - 1) / IT;
    ll nblocks = tiles * (tiles + 1) / 2;
    hipLaunchKernelGGL(k_isfc_fused, dim3((unsigned)nblocks), dim3(256),
                       0, stream, acc, (const bf16_t*)Zs,
                       (const bf16_t*)Zm, V, T, B);
}
