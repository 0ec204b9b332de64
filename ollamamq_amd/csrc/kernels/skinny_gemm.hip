// gfx950 skinny GEMM for the decode path: y[M,N] = x[M,K] @ W[N,K]^T,
// M ≤ 32 (the decode batch), bf16 in/out, fp32 MFMA accumulation.
//
// Why hand-written: decode projections are weights-streaming bound
// (N×K×2 B read once per step; x is L2-resident), and the library GEMM
// measured ~1.5-4.5 TB/s on these shapes (M=32) — far off the ≈6.3 TB/s
// HBM roofline.  This kernel maps the whole problem onto
// v_mfma_f32_32x32x16_bf16 tiles whose B-fragment is a 16-byte contiguous
// run of a W row, so every lane issues one dwordx4 per MFMA and the wave
// streams 1 KiB per instruction.
//
// Geometry: one block = 8 waves = one 32-column tile of y; wave w
// accumulates the k-segment [w*K/8, (w+1)*K/8) (in-block split-K), then
// the partials are reduced through LDS — no global partial slabs, no
// second kernel.  Grid = N/32 blocks → N/32 × 8 waves (1536 waves for the
// 8B qkv projection: enough to keep every SIMD streaming).
//
// Constraints: K % 128 == 0, N % 32 == 0, M ≤ 32.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

// A-fragment: lane l -> x[row = l&31][k0 + (l>>5)*8 .. +8)  (zero if row>=M)
// B-fragment: lane l -> W[n0 + (l&31)][k0 + (l>>5)*8 .. +8)
__global__ __launch_bounds__(512) void k_skinny_gemm(
    bf16* __restrict__ y,            // [M, N]
    const bf16* __restrict__ x,      // [M, K] row stride xs
    const bf16* __restrict__ w,      // [N, K] row-major
    int M, int N, int K, int64_t xs)
{
    const int n0 = blockIdx.x * 32;
    const int wid = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int row = lane & 31;         // x row (M) / w col (N tile)
    const int khalf = (lane >> 5) * 8;

    const int kseg = K >> 3;           // per-wave k extent
    const int k0 = wid * kseg;
    const int k1 = k0 + kseg;

    const bf16* xrow = x + (int64_t)row * xs;      // row < 32; masked below
    const bf16* wrow = w + (int64_t)(n0 + row) * K;
    const bool live_a = row < M;

    f32x16 acc = {};
    // 1-deep software prefetch; the uniform-branch-free body lets hipcc
    // keep several dwordx4 loads in flight across the MFMAs (guide §5
    // "Three .s-level traps": no per-element runtime condition on loads).
    if (live_a) {
        bf16x8 a_n = *reinterpret_cast<const bf16x8*>(xrow + k0 + khalf);
        bf16x8 b_n = *reinterpret_cast<const bf16x8*>(wrow + k0 + khalf);
        #pragma unroll 2
        for (int k = k0; k < k1 - 16; k += 16) {
            const bf16x8 a = a_n, b = b_n;
            a_n = *reinterpret_cast<const bf16x8*>(xrow + k + 16 + khalf);
            b_n = *reinterpret_cast<const bf16x8*>(wrow + k + 16 + khalf);
            acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
        }
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_n, b_n, acc, 0, 0, 0);
    } else {
        const bf16x8 a = {};
        bf16x8 b_n = *reinterpret_cast<const bf16x8*>(wrow + k0 + khalf);
        #pragma unroll 2
        for (int k = k0; k < k1 - 16; k += 16) {
            const bf16x8 b = b_n;
            b_n = *reinterpret_cast<const bf16x8*>(wrow + k + 16 + khalf);
            acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
        }
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b_n, acc, 0, 0, 0);
    }

    // ---- in-block split-K reduction through LDS ----
    // C/D layout (32x32x16): col = lane&31, row = (r&3) + 8*(r>>2) + 4*(lane>>5)
    __shared__ float red[8][32][32];
    #pragma unroll
    for (int r = 0; r < 16; r++) {
        const int crow = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        red[wid][crow][lane & 31] = acc[r];
    }
    __syncthreads();
    // 512 threads reduce 1024 outputs: 2 per thread
    const int tid = threadIdx.x;
    #pragma unroll
    for (int e = tid; e < 1024; e += 512) {
        const int m = e >> 5, n = e & 31;
        if (m < M) {
            float s = red[0][m][n] + red[1][m][n] + red[2][m][n]
                    + red[3][m][n] + red[4][m][n] + red[5][m][n]
                    + red[6][m][n] + red[7][m][n];
            y[(int64_t)m * N + n0 + n] = __float2bfloat16(s);
        }
    }
}

extern "C" int skinny_gemm_bf16(
    void* y, const void* x, const void* w, int M, int N, int K,
    int64_t xs, hipStream_t stream)
{
    k_skinny_gemm<<<N / 32, 512, 0, stream>>>(
        (bf16*)y, (const bf16*)x, (const bf16*)w, M, N, K, xs);
    return (int)hipGetLastError();
}
