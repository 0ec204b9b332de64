// gfx950 weight-streaming decode GEMM: y[M,N] = x[M,K] @ W[N,K]^T, M <= 64.
//
// The decode projections are pure weight streaming (each weight byte is
// used M<=64 times; compute is nothing next to the 8 TB/s HBM pull), and
// round-1 measured hipBLASLt at 1.8-4.6 TB/s on these shapes with a
// ~19 us floor on qkv/o (profiles/r01_decode_step_budget.md).  Two
// hand-written attempts also lost:
//  * fragment-direct (skinny_gemm.hip): MFMA-layout loads are 16 B/lane at
//    8 KB row stride -> cold misses over-fetched ~4x;
//  * LDS-staged: full-line loads, but __syncthreads() carries a vmcnt(0)
//    (cdna_hip_programming.md "Pipelining across barriers"), draining the
//    prefetch ring every chunk -> 87% SQ_WAIT_ANY.
//
// This kernel removes the conflict at the source: the weights are OURS,
// so they are stored pre-packed in MFMA fragment order (ops/hip.py
// pack_weight).  Element W[n][k] with n = t*32+r, k = b*64 + j*16 + h*8 + e
// lives at packed uint4 index ((t*NB + b)*4 + j)*64 + (h*32 + r), e inside.
// The stream is then PERFECTLY linear: each wave reads consecutive 1 KiB
// vectors (full 128 B line use per instruction), tagged non-temporal
// (MI355X_MICROARCH "nt-weights": one-time-read stream, -18% landed
// traffic), with no LDS staging, no barriers, and a register prefetch one
// 4 KiB iteration deep per wave (8 waves x ~2 blocks/CU ~= 64 KiB in
// flight/CU -> "streaming" regime, MICROARCH price-table preamble).
//
// Geometry: grid (N/32, ksplit) x 512 threads (8 waves).  The K/64
// k-blocks are range-partitioned over ksplit*8 waves (uneven ranges OK:
// any K%64==0, N%32==0, M<=64).  Wave partials meet in a single
// end-of-kernel LDS reduce (the ONE barrier, after all streaming).
// ksplit>1 writes fp32 partials; k_wstream_combine folds them (+bias).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

union u4bf8 { u32x4 u; bf16x8 v; };

template <int MT>   // m-tiles of 32 rows (1: M<=32, 2: M<=64)
__global__ __launch_bounds__(512) void k_wstream_gemm(
    bf16* __restrict__ y,            // [M, N] (ksplit == 1)
    float* __restrict__ part,        // [ksplit, M, N] (ksplit > 1)
    const bf16* __restrict__ x,      // [M, K], row stride xs
    const u32x4* __restrict__ wp,    // packed W (see header)
    const bf16* __restrict__ bias,   // [N] or null (ksplit==1 path)
    int M, int N, int K, int64_t xs, int ksplit)
{
    const int t = blockIdx.x;              // n-tile (32 cols of y)
    const int ks = blockIdx.y;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int row = lane & 31;
    const int khalf = (lane >> 5) * 8;

    const int NB = K >> 6;                 // 64-elem k-blocks
    const int GW = ksplit * 8;             // waves across k
    const int gw = ks * 8 + wid;
    const int b_lo = (int)(((int64_t)gw * NB) / GW);
    const int b_hi = (int)(((int64_t)(gw + 1) * NB) / GW);

    // clamp so every lane loads valid x rows (rows >= M never stored;
    // per-lane conditional loads would branch+vmcnt(0) per chunk)
    const int ar0 = row < M ? row : (M > 0 ? M - 1 : 0);
    const bf16* xr0 = x + (int64_t)ar0 * xs + khalf;
    const bf16* xr1 = nullptr;
    if (MT == 2) {
        const int r1 = 32 + row;
        const int ar1 = r1 < M ? r1 : (M - 1);
        xr1 = x + (int64_t)ar1 * xs + khalf;
    }

    f32x16 acc0 = {}, acc1 = {};
    if (b_lo < b_hi) {
        const u32x4* wq = wp + (((int64_t)t * NB + b_lo) * 4) * 64 + lane;
        u4bf8 b0, b1, b2, b3;
        bf16x8 a0[2], a1[2], a2[2], a3[2];
        auto lda16 = [&](bf16x8* d, int b, int off) {
            const int64_t k = ((int64_t)b << 6) + off;
            d[0] = *reinterpret_cast<const bf16x8*>(xr0 + k);
            if (MT == 2) d[1] = *reinterpret_cast<const bf16x8*>(xr1 + k);
        };
        // prologue: first iteration's 4 weight vectors + activations
        b0.u = __builtin_nontemporal_load(wq);
        b1.u = __builtin_nontemporal_load(wq + 64);
        b2.u = __builtin_nontemporal_load(wq + 128);
        b3.u = __builtin_nontemporal_load(wq + 192);
        lda16(a0, b_lo, 0);
        lda16(a1, b_lo, 16);
        lda16(a2, b_lo, 32);
        lda16(a3, b_lo, 48);
        const u32x4* wnext = wq + 256;
        for (int b = b_lo; b < b_hi - 1; ++b, wnext += 256) {
            bf16x8 vb0 = b0.v, vb1 = b1.v, vb2 = b2.v, vb3 = b3.v;
            acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                a0[0], vb0, acc0, 0, 0, 0);
            if (MT == 2)
                acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a0[1], vb0, acc1, 0, 0, 0);
            b0.u = __builtin_nontemporal_load(wnext);
            lda16(a0, b + 1, 0);
            acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                a1[0], vb1, acc0, 0, 0, 0);
            if (MT == 2)
                acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a1[1], vb1, acc1, 0, 0, 0);
            b1.u = __builtin_nontemporal_load(wnext + 64);
            lda16(a1, b + 1, 16);
            acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                a2[0], vb2, acc0, 0, 0, 0);
            if (MT == 2)
                acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a2[1], vb2, acc1, 0, 0, 0);
            b2.u = __builtin_nontemporal_load(wnext + 128);
            lda16(a2, b + 1, 32);
            acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                a3[0], vb3, acc0, 0, 0, 0);
            if (MT == 2)
                acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a3[1], vb3, acc1, 0, 0, 0);
            b3.u = __builtin_nontemporal_load(wnext + 192);
            lda16(a3, b + 1, 48);
        }
        acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0[0], b0.v, acc0,
                                                       0, 0, 0);
        acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1[0], b1.v, acc0,
                                                       0, 0, 0);
        acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a2[0], b2.v, acc0,
                                                       0, 0, 0);
        acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a3[0], b3.v, acc0,
                                                       0, 0, 0);
        if (MT == 2) {
            acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0[1], b0.v,
                                                           acc1, 0, 0, 0);
            acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1[1], b1.v,
                                                           acc1, 0, 0, 0);
            acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a2[1], b2.v,
                                                           acc1, 0, 0, 0);
            acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a3[1], b3.v,
                                                           acc1, 0, 0, 0);
        }
    }

    // ---- the one barrier: reduce 8 wave partials through LDS ----
    __shared__ float red8[8][32][32];
    const int n0 = t * 32;
    for (int mt = 0; mt < MT; ++mt) {
        const f32x16& acc = mt ? acc1 : acc0;
        if (mt) __syncthreads();
        #pragma unroll
        for (int r = 0; r < 16; r++) {
            const int crow = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
            red8[wid][crow][lane & 31] = acc[r];
        }
        __syncthreads();
        #pragma unroll
        for (int e = tid; e < 1024; e += 512) {
            const int m = (e >> 5) + mt * 32, n = e & 31;
            if (m < M) {
                float s = 0.f;
                #pragma unroll
                for (int wv = 0; wv < 8; wv++)
                    s += red8[wv][m - mt * 32][n];
                if (ksplit == 1) {
                    if (bias) s += __bfloat162float(bias[n0 + n]);
                    y[(int64_t)m * N + n0 + n] = __float2bfloat16(s);
                } else {
                    part[((int64_t)ks * M + m) * N + n0 + n] = s;
                }
            }
        }
    }
}

__global__ __launch_bounds__(256) void k_wstream_combine(
    bf16* __restrict__ y, const float* __restrict__ part,
    const bf16* __restrict__ bias, int64_t total, int64_t mn, int N,
    int ksplit)
{
    for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        float s = 0.f;
        for (int k = 0; k < ksplit; k++) s += part[k * mn + i];
        if (bias) s += __bfloat162float(bias[i % N]);
        y[i] = __float2bfloat16(s);
    }
}

extern "C" int wstream_gemm_bf16(
    void* y, void* part, const void* x, const void* wp, const void* bias,
    int M, int N, int K, int64_t xs, int ksplit, hipStream_t stream)
{
    dim3 grid(N / 32, ksplit);
    if (M <= 32)
        k_wstream_gemm<1><<<grid, 512, 0, stream>>>(
            (bf16*)y, (float*)part, (const bf16*)x, (const u32x4*)wp,
            (const bf16*)bias, M, N, K, xs, ksplit);
    else
        k_wstream_gemm<2><<<grid, 512, 0, stream>>>(
            (bf16*)y, (float*)part, (const bf16*)x, (const u32x4*)wp,
            (const bf16*)bias, M, N, K, xs, ksplit);
    if (ksplit > 1) {
        const int64_t mn = (int64_t)M * N;
        const int64_t want = (mn + 255) / 256;
        const int blocks = (int)(want < 1024 ? want : 1024);
        k_wstream_combine<<<blocks, 256, 0, stream>>>(
            (bf16*)y, (const float*)part, (const bf16*)bias, mn, mn, N,
            ksplit);
    }
    return (int)hipGetLastError();
}
