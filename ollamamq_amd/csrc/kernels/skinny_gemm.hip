// gfx950 skinny GEMM for the decode path: y[M,N] = x[M,K] @ W[N,K]^T,
// M ≤ 32 (decode batch), bf16 in/out, fp32 MFMA accumulation.
//
// STATUS (measured, tools/perf_gemm.py + bench.py A/B): kept in-tree as
// opt-in (OLLAMAMQ_SKINNY_MAX_N/K env gates; default = library GEMMs).
// Two variants below:
//  * k_skinny_gemm ("staged"): W chunks staged through LDS in full
//    coalesced 128 B lines, 3-buffer ring, 2-deep register prefetch,
//    in-block + grid k-split.  ~2 TB/s — stall-bound (PMC: 87%
//    SQ_WAIT_ANY at 1 block/CU before the ring; the ring recovered only
//    part of it).
//  * k_skinny_direct: fragment-direct (no LDS, no barriers), 4-deep
//    register prefetch, 8 independent waves/block + grid k-split.
//    Beats hipBLASLt on L3-WARM microbenches (qkv 2.9 vs 2.7 TB/s,
//    o 2.4 vs 1.8) but loses ~0.3 ms/step in the real serving loop where
//    weights stream COLD from HBM: its 16 B-per-lane loads at 8 KB row
//    stride use 32 B per fetched 128 B line per pass, so cold misses
//    over-fetch ~4x (the L3 hid this in the microbench).  Honest A/B in
//    bench.py decided the default (5473 lib vs 5221 direct tok/s).
//
// Constraints: K % 256 == 0 (staged) / K % (8*64*ksplit) == 0 (direct),
// N % 32 == 0, M ≤ 32.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#define KCH 128                  // staged k elems per chunk
#define BN 32                    // output columns per block
#define LROW (KCH + 8)           // LDS row stride (bf16), +16 B pad
#define NBUF 3                   // LDS ring depth (2-deep load prefetch)

// grid (N/BN, ksplit); block ks owns k-range [ks*K/ksplit, ...).
// ksplit>1 writes fp32 partials to part[ks][M][N]; k_skinny_combine sums.
__global__ __launch_bounds__(256) void k_skinny_gemm(
    bf16* __restrict__ y,            // [M, N] (ksplit == 1)
    float* __restrict__ part,        // [ksplit, M, N] (ksplit > 1)
    const bf16* __restrict__ x,      // [M, K], row stride xs
    const bf16* __restrict__ w,      // [N, K] row-major
    int M, int N, int K, int64_t xs, int ksplit)
{
    const int n0 = blockIdx.x * BN;
    const int ks = blockIdx.y;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int row = lane & 31;       // x row (M) / fragment row
    const int khalf = (lane >> 5) * 8;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    bf16* wt = reinterpret_cast<bf16*>(smem);      // [NBUF][BN][LROW]
    float* red = reinterpret_cast<float*>(wt + NBUF * BN * LROW);

    // staging geometry: 8 KiB chunk = BN rows × 256 B; thread t covers
    // (row = t>>3, 16 B at (t&7)*16) × 2 rounds of 128 B — consecutive
    // threads hit consecutive 16 B of one row: full-line coalescing.
    const int s_row = tid >> 3;
    const int s_off = (tid & 7) * 16;
    const int64_t wbase = (int64_t)(n0 + s_row) * K;

    const int kseg = K / ksplit;          // this block's k extent
    const int k0 = ks * kseg;
    const int nch = kseg / KCH;           // hip.py guarantees even, >= 2
    // clamp the row so EVERY lane loads valid memory: a per-lane
    // conditional load makes hipcc branch around each load with a
    // vmcnt(0) (guide §5 traps (c): +11k cycles/block).  C rows >= M are
    // duplicates of row M-1 and are never stored.
    const int arow = row < M ? row : (M - 1);
    const bf16* xrow = x + (int64_t)arow * xs;

    // 2-deep staging prefetch: two register sets (sA even chunks, sB odd)
    // and an LDS ring of 3 — a chunk's loads stay in flight across a FULL
    // loop iteration (the depth-1 version exposed ~the whole HBM latency
    // per chunk: PMC showed 87% SQ_WAIT_ANY).
    uint4 sA[2], sB[2];
    auto load_to = [&](uint4* dst, int c) {
        const int64_t base = wbase + k0 + (int64_t)c * KCH;  // elems
        #pragma unroll
        for (int r = 0; r < 2; r++)
            dst[r] = *reinterpret_cast<const uint4*>(
                reinterpret_cast<const char*>(w + base) + r * 128 + s_off);
    };
    auto write_from = [&](const uint4* src, int buf) {
        bf16* dst = wt + buf * BN * LROW + s_row * LROW;
        #pragma unroll
        for (int r = 0; r < 2; r++)
            *reinterpret_cast<uint4*>(
                reinterpret_cast<char*>(dst) + r * 128 + s_off) = src[r];
    };

    f32x16 acc = {};
    const int cmax = nch - 1;
    load_to(sA, 0);
    write_from(sA, 0);
    load_to(sB, 1 < cmax ? 1 : cmax);
    __syncthreads();

    for (int c = 0; c < nch; c += 2) {
        // --- even chunk c: compute buf[c%3], stage chunk c+1 (sB) into
        //     buf[(c+1)%3], refill sA with chunk c+2 ---
        {
            const bf16* xk = xrow + k0 + (int64_t)c * KCH + wid * 32 + khalf;
            bf16x8 a[2];
            a[0] = *reinterpret_cast<const bf16x8*>(xk);
            a[1] = *reinterpret_cast<const bf16x8*>(xk + 16);
            load_to(sA, c + 2 < cmax ? c + 2 : cmax);
            const bf16* wrow = wt + (c % NBUF) * BN * LROW + row * LROW
                               + wid * 32 + khalf;
            const bf16x8 b0 = *reinterpret_cast<const bf16x8*>(wrow);
            const bf16x8 b1 = *reinterpret_cast<const bf16x8*>(wrow + 16);
            acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a[0], b0, acc,
                                                          0, 0, 0);
            acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a[1], b1, acc,
                                                          0, 0, 0);
            __syncthreads();
            write_from(sB, (c + 1) % NBUF);
            __syncthreads();
        }
        // --- odd chunk c+1: compute buf[(c+1)%3], stage chunk c+2 (sA)
        //     into buf[(c+2)%3], refill sB with chunk c+3 ---
        if (c + 1 < nch) {
            const int c1 = c + 1;
            const bf16* xk = xrow + k0 + (int64_t)c1 * KCH + wid * 32
                             + khalf;
            bf16x8 a[2];
            a[0] = *reinterpret_cast<const bf16x8*>(xk);
            a[1] = *reinterpret_cast<const bf16x8*>(xk + 16);
            load_to(sB, c1 + 2 < cmax ? c1 + 2 : cmax);
            const bf16* wrow = wt + (c1 % NBUF) * BN * LROW + row * LROW
                               + wid * 32 + khalf;
            const bf16x8 b0 = *reinterpret_cast<const bf16x8*>(wrow);
            const bf16x8 b1 = *reinterpret_cast<const bf16x8*>(wrow + 16);
            acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a[0], b0, acc,
                                                          0, 0, 0);
            acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a[1], b1, acc,
                                                          0, 0, 0);
            __syncthreads();
            write_from(sA, (c1 + 1) % NBUF);
            __syncthreads();
        }
    }

    // ---- reduce the 4 wave partials through LDS ----
    // C/D layout (32x32x16): col=lane&31, row=(r&3)+8*(r>>2)+4*(lane>>5)
    #pragma unroll
    for (int r = 0; r < 16; r++) {
        const int crow = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        red[(wid * 32 + crow) * 32 + (lane & 31)] = acc[r];
    }
    __syncthreads();
    #pragma unroll
    for (int e = tid; e < 1024; e += 256) {
        const int m = e >> 5, n = e & 31;
        if (m < M) {
            const float s = red[(0 * 32 + m) * 32 + n]
                          + red[(1 * 32 + m) * 32 + n]
                          + red[(2 * 32 + m) * 32 + n]
                          + red[(3 * 32 + m) * 32 + n];
            if (ksplit == 1)
                y[(int64_t)m * N + n0 + n] = __float2bfloat16(s);
            else
                part[((int64_t)ks * M + m) * N + n0 + n] = s;
        }
    }
}

__global__ __launch_bounds__(256) void k_skinny_combine(
    bf16* __restrict__ y, const float* __restrict__ part,
    int64_t total, int64_t mn, int ksplit)
{
    for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        float s = 0.f;
        for (int k = 0; k < ksplit; k++) s += part[k * mn + i];
        y[i] = __float2bfloat16(s);
    }
}

extern "C" int skinny_gemm_bf16(
    void* y, void* part, const void* x, const void* w, int M, int N, int K,
    int64_t xs, int ksplit, hipStream_t stream)
{
    const int lds = NBUF * BN * LROW * 2 + 4 * 32 * 32 * 4;
    dim3 grid(N / BN, ksplit);
    k_skinny_gemm<<<grid, 256, lds, stream>>>(
        (bf16*)y, (float*)part, (const bf16*)x, (const bf16*)w, M, N, K,
        xs, ksplit);
    if (ksplit > 1) {
        const int64_t mn = (int64_t)M * N;
        const int64_t want = (mn + 255) / 256;
        const int blocks = (int)(want < 1024 ? want : 1024);
        k_skinny_combine<<<blocks, 256, 0, stream>>>(
            (bf16*)y, (const float*)part, mn, mn, ksplit);
    }
    return (int)hipGetLastError();
}

// ---------------------------------------------------------------------
// Fragment-direct variant: no LDS staging, no barriers — every wave
// streams its own (32-col tile, k-slice) with a 4-deep register
// prefetch ring, so latency is hidden by wave count alone (grid k-split
// pushes occupancy to ~16+ waves/CU).  B loads are 16 B/lane at 8 KB row
// stride; rows are re-touched every k-step so L1/L2 serve the interior
// of each 128 B line.
// grid (N/BN, ksplit); block = 8 waves, wave w takes k-slice w of 8.
__global__ __launch_bounds__(512) void k_skinny_direct(
    bf16* __restrict__ y, float* __restrict__ part,
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    int M, int N, int K, int64_t xs, int ksplit)
{
    const int n0 = blockIdx.x * BN;
    const int ks = blockIdx.y;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int row = lane & 31;
    const int khalf = (lane >> 5) * 8;

    const int kseg = K / (ksplit * 8);       // per-wave k extent
    const int k0 = (ks * 8 + wid) * kseg;
    const int k1 = k0 + kseg;                // kseg % 64 == 0 (hip.py)

    const int arow = row < M ? row : (M - 1);
    const bf16* xrow = x + (int64_t)arow * xs + khalf;
    const bf16* wrow = w + (int64_t)(n0 + row) * K + khalf;

    f32x16 acc = {};
    bf16x8 a0, a1, a2, a3, b0, b1, b2, b3;
    a0 = *reinterpret_cast<const bf16x8*>(xrow + k0);
    b0 = *reinterpret_cast<const bf16x8*>(wrow + k0);
    a1 = *reinterpret_cast<const bf16x8*>(xrow + k0 + 16);
    b1 = *reinterpret_cast<const bf16x8*>(wrow + k0 + 16);
    a2 = *reinterpret_cast<const bf16x8*>(xrow + k0 + 32);
    b2 = *reinterpret_cast<const bf16x8*>(wrow + k0 + 32);
    a3 = *reinterpret_cast<const bf16x8*>(xrow + k0 + 48);
    b3 = *reinterpret_cast<const bf16x8*>(wrow + k0 + 48);
    for (int k = k0; k < k1 - 64; k += 64) {
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc, 0, 0, 0);
        a0 = *reinterpret_cast<const bf16x8*>(xrow + k + 64);
        b0 = *reinterpret_cast<const bf16x8*>(wrow + k + 64);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc, 0, 0, 0);
        a1 = *reinterpret_cast<const bf16x8*>(xrow + k + 80);
        b1 = *reinterpret_cast<const bf16x8*>(wrow + k + 80);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a2, b2, acc, 0, 0, 0);
        a2 = *reinterpret_cast<const bf16x8*>(xrow + k + 96);
        b2 = *reinterpret_cast<const bf16x8*>(wrow + k + 96);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a3, b3, acc, 0, 0, 0);
        a3 = *reinterpret_cast<const bf16x8*>(xrow + k + 112);
        b3 = *reinterpret_cast<const bf16x8*>(wrow + k + 112);
    }
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a2, b2, acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a3, b3, acc, 0, 0, 0);

    __shared__ float red8[8][32][32];
    #pragma unroll
    for (int r = 0; r < 16; r++) {
        const int crow = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        red8[wid][crow][lane & 31] = acc[r];
    }
    __syncthreads();
    #pragma unroll
    for (int e = tid; e < 1024; e += 512) {
        const int m = e >> 5, n = e & 31;
        if (m < M) {
            float s = 0.f;
            #pragma unroll
            for (int wv = 0; wv < 8; wv++) s += red8[wv][m][n];
            if (ksplit == 1)
                y[(int64_t)m * N + n0 + n] = __float2bfloat16(s);
            else
                part[((int64_t)ks * M + m) * N + n0 + n] = s;
        }
    }
}

extern "C" int skinny_direct_bf16(
    void* y, void* part, const void* x, const void* w, int M, int N, int K,
    int64_t xs, int ksplit, hipStream_t stream)
{
    dim3 grid(N / BN, ksplit);
    k_skinny_direct<<<grid, 512, 0, stream>>>(
        (bf16*)y, (float*)part, (const bf16*)x, (const bf16*)w, M, N, K,
        xs, ksplit);
    if (ksplit > 1) {
        const int64_t mn = (int64_t)M * N;
        const int64_t want = (mn + 255) / 256;
        const int blocks = (int)(want < 1024 ? want : 1024);
        k_skinny_combine<<<blocks, 256, 0, stream>>>(
            (bf16*)y, (const float*)part, mn, mn, ksplit);
    }
    return (int)hipGetLastError();
}
