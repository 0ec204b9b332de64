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
DEV float bf2f(bf16 x) { return __bfloat162float(x); }

#define PF_CHUNK 64          // kv tokens per staged chunk
#define PF_ROWS 32           // q rows per tile
#define DH 128
#define KTROW (DH + 4)       // k_tile row stride (bf16)
#define VTROW (PF_CHUNK + 8) // v_t row stride (bf16): [128 d][64 kv]
#define PROW (PF_CHUNK + 8)  // p_lds row stride

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
    int Hq, int KVH, int page, int max_pages, float scale, int64_t qs)
{
    const int tile = blockIdx.x;
    const int kvh = blockIdx.y;
    const int G = Hq / KVH;
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
    bf16* k_tile = reinterpret_cast<bf16*>(smem);              // [64][KTROW]
    bf16* v_t = k_tile + PF_CHUNK * KTROW;                     // [128][VTROW]
    bf16* p_lds = v_t + DH * VTROW;       // [G][32][PROW]
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

    const int n_chunks = (kv_len + PF_CHUNK - 1) / PF_CHUNK;
    for (int ch = 0; ch < n_chunks; ch++) {
        const int base = ch * PF_CHUNK;
        const int n_here = min(PF_CHUNK, kv_len - base);
        __syncthreads();
        // ---- stage K (row-major) and V (transposed) ----
        {
            const int tid = threadIdx.x, nthr = blockDim.x;
            for (int u = tid; u < PF_CHUNK * (DH / 8); u += nthr) {
                const int tok = u / (DH / 8), dv = u % (DH / 8);
                // clamped, unconditional loads (guide §5 traps (c));
                // dead kv rows are causally masked to p = 0 below
                const int tk = base + tok < kv_len ? base + tok
                                                   : kv_len - 1;
                const int gp = page_table[(int64_t)slot * max_pages
                                          + tk / page];
                const int64_t src = (((int64_t)gp * KVH + kvh) * page
                                     + tk % page) * DH + dv * 8;
                const uint4 kv4 = *reinterpret_cast<const uint4*>(kpool + src);
                const uint4 vv4 = *reinterpret_cast<const uint4*>(vpool + src);
                *reinterpret_cast<uint4*>(k_tile + tok * KTROW + dv * 8) = kv4;
                const bf16* vsrc = reinterpret_cast<const bf16*>(&vv4);
                #pragma unroll
                for (int e = 0; e < 8; e++)
                    v_t[(dv * 8 + e) * VTROW + tok] = vsrc[e];
            }
        }
        __syncthreads();

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
        #pragma unroll
        for (int t = 0; t < 2; t++)
            #pragma unroll
            for (int r = 0; r < 16; r++) {
                const int kv = base + t * 32 + (r & 3) + 8 * (r >> 2)
                               + 4 * (lane >> 5);
                const bool ok = kv < kv_len && kv <= qpos && col < rows;
                s[t][r] = ok ? s[t][r] * scale : -3.0e38f;
                pmax = fmaxf(pmax, s[t][r]);
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
            const bf16* vrow = v_t + (t * 32 + col) * VTROW + khalf;
            const bf16* prow = my_p + col * PROW + khalf;
            #pragma unroll
            for (int c = 0; c < 4; c++) {
                const bf16x8 a =
                    *reinterpret_cast<const bf16x8*>(vrow + c * 16);
                const bf16x8 b =
                    *reinterpret_cast<const bf16x8*>(prow + c * 16);
                o[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a, b, o[t], 0, 0, 0);
            }
        }
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
    hipStream_t stream)
{
    const int G = Hq / KVH;
    const int lds = PF_CHUNK * KTROW * 2 + DH * VTROW * 2
                    + G * PF_ROWS * PROW * 2;
    static int allowed = 0;
    if (!allowed && lds > 64 * 1024) {
        (void)hipFuncSetAttribute(
            (const void*)k_prefill_attn,
            hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        allowed = 1;
    }
    dim3 grid(n_tiles, KVH);
    k_prefill_attn<<<grid, G * 64, lds, stream>>>(
        (bf16*)out, (const bf16*)q, (const bf16*)kpool, (const bf16*)vpool,
        (const int*)page_table, (const int*)tile_slot,
        (const int*)tile_q0, (const int*)tile_pos0, (const int*)tile_rows,
        Hq, KVH, page, max_pages, scale, q_stride);
    return (int)hipGetLastError();
}
