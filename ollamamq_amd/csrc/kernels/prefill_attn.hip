// gfx950 MFMA prefill attention over the paged KV pool (flash-style,
// causal, GQA, varlen tiles) — replaces the VALU prefill path (measured
// 3.4 ms per 4096-token layer; QK^T/PV belong on the matrix cores).
//
// Geometry: grid (n_tiles, KVH); block = G waves (G = Hq/KVH ≤ 8); wave w
// computes q-head (kvh*G + w) for the tile's 32 query rows.  Per 64-token
// KV chunk, staged once into LDS and shared by the group's G heads:
//
//   S^T[64kv, 32q] = mfma(A=K,  B=Q)   "swapped QK^T" (guide §B): the C
//       layout (col = lane&31) makes q per-LANE, so the online softmax
//       (m, l) is 2 scalars per lane and the kv-reduce is in-register
//       (15 adds + one permlane32 half-swap), no cross-lane trees;
//   P -> p_lds[32q][64kv] bf16;
//   O^T[128d, 32q] += mfma(A=V^T, B=P) over 4 d-tiles — the LDS V image
//       is stored TRANSPOSED at staging time so both PV fragments are
//       contiguous 16-byte ds reads.
//
// A/B/C fragment maps (verified numerically by the skinny-GEMM tests):
//   A: row = lane&31,  k = (lane>>5)*8 + j
//   B: col = lane&31,  k = (lane>>5)*8 + j
//   C: col = lane&31, row = (r&3) + 8*(r>>2) + 4*(lane>>5)
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __hip_bfloat16 bf16;
typedef __hip_bfloat162 bf162;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#define DEV static __device__ __forceinline__
[[maybe_unused]] DEV float bf2f(bf16 x) { return __bfloat162float(x); }

#define PF_CHUNK 64          // kv tokens per staged chunk
#define PF_ROWS 32           // q rows per tile
#define DH 128
#define KTROW (DH + 8)       // k_tile row stride: 272 B rows stay 16-B
                             // aligned for ds_read_b128 (G17) and the
                             // 68-dword stride is conflict-free per group
#define VTROW (PF_CHUNK + 8) // v_t row stride (bf16): [128 d][64 kv]
#define PROW (PF_CHUNK + 8)  // p_lds row stride
// v_t kv-BLOCK swizzle: the transposed V writes put every lane (d stride
// 8) on one bank (8*VTROW/2 ≡ 0 mod 32 dwords → measured 42% of LDS
// cycles were conflicts).  XORing the 8-element kv block index with
// (d>>3)&7 spreads the banks; reads stay 16-B contiguous per block.
#define VSWZ(d, kv) ((kv & 7) + 8 * (((kv) >> 3) ^ (((d) >> 3) & 7)))

template <int G>
__global__ __launch_bounds__(512) void k_prefill_attn(
    bf16* __restrict__ out,           // [T, Hq, D]
    const bf16* __restrict__ q,       // [T, Hq, D], row stride qs
    const bf16* __restrict__ kpool,   // [P][KVH][page][D]
    const bf16* __restrict__ vpool,
    const int* __restrict__ page_table,
    const int* __restrict__ tile_slot,
    const int* __restrict__ tile_q0,
    const int* __restrict__ tile_pos0,
    const int* __restrict__ tile_rows,
    int Hq, int KVH, int page, int max_pages, float scale, int64_t qs,
    int window)                           // 0 = full causal
{
    const int tile = blockIdx.x;
    const int kvh = blockIdx.y;
    const int wid = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int qh = kvh * G + wid;
    const int col = lane & 31;          // this lane's q row within the tile
    const int khalf = (lane >> 5) * 8;

    const int slot = tile_slot[tile];
    const int q0 = tile_q0[tile];
    const int pos0 = tile_pos0[tile];
    const int rows = tile_rows[tile];
    const int kv_len = pos0 + rows;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    // double-buffered K/V chunk tiles (T14: next chunk's loads issued
    // before this chunk's MFMAs, written after the barrier)
    bf16* k_tile0 = reinterpret_cast<bf16*>(smem);             // [64][KTROW]
    bf16* v_t0 = k_tile0 + 2 * PF_CHUNK * KTROW;               // [128][VTROW]
    bf16* p_lds = v_t0 + 2 * DH * VTROW;  // [G][32][PROW]
    bf16* my_p = p_lds + (int64_t)wid * PF_ROWS * PROW;

    // ---- load this wave's Q fragments into registers ----
    // B-operand of QK^T: lane needs Q[q=col][k 8-contig at khalf + 16*c]
    bf16x8 qf[8];
    {
        // clamped row: dead q columns are masked in the softmax and
        // never stored (conditional loads de-pipeline: traps (c))
        const int qrow_i = col < rows ? col : (rows - 1);
        const bf16* qrow = q + (int64_t)(q0 + qrow_i) * qs
                           + (int64_t)qh * DH;
        #pragma unroll
        for (int c = 0; c < 8; c++)
            qf[c] = *reinterpret_cast<const bf16x8*>(qrow + c * 16 + khalf);
    }

    // ---- per-lane online softmax state (q = col) ----
    float m = -3.0e38f, l = 0.f;
    // O^T accumulators: 4 d-tiles of [32d x 32q]
    f32x16 o[4] = {};

    // staging geometry: each thread owns N_PIECES 16-B pieces of the
    // chunk; loads for chunk c+1 are issued before chunk c's MFMAs and
    // written to the other buffer after the barrier.  Compile-time G
    // keeps the piece arrays in registers (runtime indexing would go to
    // scratch — guide §5.4 rule 20).
    constexpr int NTHR = G * 64;
    constexpr int NSLOT = PF_CHUNK * (DH / 8);
    constexpr int N_PIECES = (NSLOT + NTHR - 1) / NTHR;
    // odd GQA groups (G=7, Qwen2): NSLOT % NTHR != 0 — clamp the last
    // piece's slot; the duplicate threads re-load/re-write identical data
    // (a guarded load/write would de-pipeline the stage, traps (c))
    uint4 stK[N_PIECES], stV[N_PIECES];
    const int tid = threadIdx.x;

    auto issue_chunk = [&](int base) {
        #pragma unroll
        for (int pi = 0; pi < N_PIECES; pi++) {
            const int u = min(tid + pi * NTHR, NSLOT - 1);
            const int tok = u / (DH / 8);
            const int dv = u % (DH / 8);
            const int tk = base + tok < kv_len ? base + tok : kv_len - 1;
            const int gp = page_table[(int64_t)slot * max_pages
                                      + tk / page];
            const int64_t src = (((int64_t)gp * KVH + kvh) * page
                                 + tk % page) * DH + dv * 8;
            stK[pi] = *reinterpret_cast<const uint4*>(kpool + src);
            stV[pi] = *reinterpret_cast<const uint4*>(vpool + src);
        }
    };
    auto write_chunk = [&](int buf) {
        bf16* kt = k_tile0 + buf * PF_CHUNK * KTROW;
        bf16* vt = v_t0 + buf * DH * VTROW;
        #pragma unroll
        for (int pi = 0; pi < N_PIECES; pi++) {
            const int u = min(tid + pi * NTHR, NSLOT - 1);
            const int tok = u / (DH / 8);
            const int dv = u % (DH / 8);
            *reinterpret_cast<uint4*>(kt + tok * KTROW + dv * 8) = stK[pi];
            const bf16* vsrc = reinterpret_cast<const bf16*>(&stV[pi]);
            #pragma unroll
            for (int e = 0; e < 8; e++) {
                const int d = dv * 8 + e;
                vt[d * VTROW + VSWZ(d, tok)] = vsrc[e];
            }
        }
    };

    const int n_chunks = (kv_len + PF_CHUNK - 1) / PF_CHUNK;
    // sliding window: the earliest key any q row of this tile can see is
    // pos0 - window + 1 (row 0); skip chunks entirely below it
    const int ch_lo = (window > 0 && pos0 >= window)
                          ? (pos0 - window + 1) / PF_CHUNK : 0;
    issue_chunk(ch_lo * PF_CHUNK);
    write_chunk(ch_lo & 1);
    __syncthreads();
    for (int ch = ch_lo; ch < n_chunks; ch++) {
        const int base = ch * PF_CHUNK;
        const int n_here = min(PF_CHUNK, kv_len - base);
        (void)n_here;
        bf16* k_tile = k_tile0 + (ch & 1) * PF_CHUNK * KTROW;
        bf16* v_t = v_t0 + (ch & 1) * DH * VTROW;
        // issue next chunk's loads (clamped on the last chunk)
        issue_chunk(ch + 1 < n_chunks ? (ch + 1) * PF_CHUNK : base);

        // ---- S^T[64kv, 32q] = K x Q^T, two 32-kv tiles ----
        f32x16 s[2] = {};
        #pragma unroll
        for (int t = 0; t < 2; t++) {
            const bf16* krow = k_tile + (t * 32 + col) * KTROW + khalf;
            #pragma unroll
            for (int c = 0; c < 8; c++) {
                const bf16x8 a =
                    *reinterpret_cast<const bf16x8*>(krow + c * 16);
                s[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a, qf[c], s[t], 0, 0, 0);
            }
        }

        // ---- causal mask + per-lane online softmax (q = col) ----
        const int qpos = pos0 + col;        // absolute position of this q
        float pmax = -3.0e38f;
        // interior chunks (base+63 <= every row's qpos, no tail) need no
        // per-element masking — the mask VALU was ~1/3 of the kernel's
        // non-MFMA instructions (uniform branch: fine)
        const bool interior = (base + PF_CHUNK <= pos0 + 1)
                              && (base + PF_CHUNK <= kv_len)
                              && (rows == PF_ROWS)
                              && (window <= 0
                                  || base >= pos0 + rows - window);
        if (interior) {
            #pragma unroll
            for (int t = 0; t < 2; t++)
                #pragma unroll
                for (int r = 0; r < 16; r++) {
                    s[t][r] *= scale;
                    pmax = fmaxf(pmax, s[t][r]);
                }
        } else {
            #pragma unroll
            for (int t = 0; t < 2; t++)
                #pragma unroll
                for (int r = 0; r < 16; r++) {
                    const int kv = base + t * 32 + (r & 3) + 8 * (r >> 2)
                                   + 4 * (lane >> 5);
                    const bool ok = kv < kv_len && kv <= qpos && col < rows
                                    && (window <= 0 || kv > qpos - window);
                    s[t][r] = ok ? s[t][r] * scale : -3.0e38f;
                    pmax = fmaxf(pmax, s[t][r]);
                }
        }
        // lanes l and l+32 each hold HALF of q-row (l&31)'s kv scores
        // (the C layout's 4*(lane>>5) row offset): combine the pair's
        // running max and sum so both halves share one softmax state
        pmax = fmaxf(pmax, __shfl_xor(pmax, 32, 64));
        const float mn = fmaxf(m, pmax);
        float corr = 1.f, psum = 0.f;
        if (mn > -3.0e38f) {
            corr = __expf(m - mn);
            #pragma unroll
            for (int t = 0; t < 2; t++)
                #pragma unroll
                for (int r = 0; r < 16; r++) {
                    const float p = (s[t][r] > -3.0e38f)
                                        ? __expf(s[t][r] - mn) : 0.f;
                    s[t][r] = p;
                    psum += p;
                }
            psum += __shfl_xor(psum, 32, 64);
            m = mn;
        } else {
            #pragma unroll
            for (int t = 0; t < 2; t++)
                #pragma unroll
                for (int r = 0; r < 16; r++) s[t][r] = 0.f;
        }
        l = l * corr + psum;
        #pragma unroll
        for (int t = 0; t < 4; t++)
            #pragma unroll
            for (int r = 0; r < 16; r++) o[t][r] *= corr;

        // ---- P -> p_lds[q = col][kv], bf16 ----
        #pragma unroll
        for (int t = 0; t < 2; t++)
            #pragma unroll
            for (int r = 0; r < 16; r++) {
                const int kv = t * 32 + (r & 3) + 8 * (r >> 2)
                               + 4 * (lane >> 5);
                my_p[col * PROW + kv] = __float2bfloat16(s[t][r]);
            }
        // p_lds is wave-private (my_p): in-wave LDS ordering suffices

        // ---- O^T[d, q] += V^T x P : 4 d-tiles, kv = 64 ----
        #pragma unroll
        for (int t = 0; t < 4; t++) {
            const int d = t * 32 + col;
            const bf16* vrow = v_t + d * VTROW;
            const bf16* prow = my_p + col * PROW + khalf;
            #pragma unroll
            for (int c = 0; c < 4; c++) {
                const int kv0 = c * 16 + khalf;   // 8-aligned block base
                const bf16x8 a = *reinterpret_cast<const bf16x8*>(
                    vrow + VSWZ(d, kv0));
                const bf16x8 b =
                    *reinterpret_cast<const bf16x8*>(prow + c * 16);
                o[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a, b, o[t], 0, 0, 0);
            }
        }
        __syncthreads();
        if (ch + 1 < n_chunks) write_chunk((ch + 1) & 1);
        __syncthreads();
    }

    // ---- epilogue: normalize and write O (lane's q = col) ----
    // o[t][r] = O^T[d = t*32 + crow(r)][q = col] => per lane a strided
    // d-column of its q row; write directly (16-bit stores).
    const float linv = l > 0.f ? 1.f / l : 0.f;
    if (col < rows) {
        bf16* orow = out + (int64_t)(q0 + col) * Hq * DH + (int64_t)qh * DH;
        #pragma unroll
        for (int t = 0; t < 4; t++)
            #pragma unroll
            for (int r = 0; r < 16; r++) {
                const int d = t * 32 + (r & 3) + 8 * (r >> 2)
                              + 4 * (lane >> 5);
                orow[d] = __float2bfloat16(o[t][r] * linv);
            }
    }
}

extern "C" int prefill_attn_bf16(
    void* out, const void* q, const void* kpool, const void* vpool,
    const void* page_table, const void* tile_slot, const void* tile_q0,
    const void* tile_pos0, const void* tile_rows, int n_tiles,
    int Hq, int KVH, int page, int max_pages, float scale, int64_t q_stride,
    int window, hipStream_t stream)
{
    const int G = Hq / KVH;
    const int lds = 2 * PF_CHUNK * KTROW * 2 + 2 * DH * VTROW * 2
                    + G * PF_ROWS * PROW * 2;
    dim3 grid(n_tiles, KVH);
    #define PF_LAUNCH(GG)                                                  \
        do {                                                               \
            static int allowed_##GG = 0;                                   \
            if (!allowed_##GG && lds > 64 * 1024) {                        \
                (void)hipFuncSetAttribute(                                 \
                    (const void*)k_prefill_attn<GG>,                       \
                    hipFuncAttributeMaxDynamicSharedMemorySize,            \
                    160 * 1024);                                           \
                allowed_##GG = 1;                                          \
            }                                                              \
            k_prefill_attn<GG><<<grid, GG * 64, lds, stream>>>(            \
                (bf16*)out, (const bf16*)q, (const bf16*)kpool,            \
                (const bf16*)vpool, (const int*)page_table,                \
                (const int*)tile_slot, (const int*)tile_q0,                \
                (const int*)tile_pos0, (const int*)tile_rows,              \
                Hq, KVH, page, max_pages, scale, q_stride, window);        \
        } while (0)
    switch (G) {
        case 3: PF_LAUNCH(3); break;
        case 5: PF_LAUNCH(5); break;
        case 6: PF_LAUNCH(6); break;
        case 7: PF_LAUNCH(7); break;
        case 1: PF_LAUNCH(1); break;
        case 2: PF_LAUNCH(2); break;
        case 4: PF_LAUNCH(4); break;
        case 8: PF_LAUNCH(8); break;
        default: return (int)hipErrorInvalidValue;
    }
    #undef PF_LAUNCH
    return (int)hipGetLastError();
}
