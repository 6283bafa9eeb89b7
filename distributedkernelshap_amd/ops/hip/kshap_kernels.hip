// MI355X (gfx950) CDNA4 kernels for distributed KernelSHAP.
//
// Native implementation of the numeric inner loop the reference delegates to
// shap 0.35.0 (reference call site explainers/kernel_shap.py:250; kernel
// inventory SURVEY.md §2.4):
//   K2  fill_random_masks       — counter-based (Philox4x32-10) coalition
//                                 sampling, complement-paired; 8 waves per
//                                 instance with ballot/prefix row allocation
//   K2b pack_masks              — u8 mask rows -> packed u64 bits for the
//                                 MFMA WLS Gram build
//   K3/K4/K5/K6 fused_predict_linear / fused_predict_bf16 —
//                                 MFMA-tiled mask @ diff GEMM
//                                 (v_mfma_f32_16x16x4_f32, or
//                                 v_mfma_f32_16x16x32_bf16 with an optional
//                                 hi+lo split B for fp32-grade results) with
//                                 LDS-staged diff tiles, the A operand
//                                 converted in-register from the raw u8
//                                 masks, fused activation (none / sigmoid /
//                                 softmax / binary-softmax-from-logit-
//                                 difference) and fused weighted background
//                                 reduction -> ey.  The masked-background
//                                 perturbation synthesis is folded
//                                 algebraically: for a linear predictor,
//                                 logits(synth[s,n]) = base[n] + sum_g
//                                 mask[s,g]*(x_part[g]-bg_part[n,g]), so the
//                                 207,200-row synth matrix never touches HBM.
//   build_diff_f32 / build_diff_bf16 — diff images written directly in the
//                                 fused kernels' operand layouts.
//   K3' synth_chunk             — explicit masked-background blend for the
//                                 arbitrary-(torch)-predictor path.
//   K7  wls_solve_mfma / wls_solve — batched constrained weighted-least-
//                                 squares: Gram+rhs built on matrix cores
//                                 (one 16x16x4 MFMA per 4 samples computes
//                                 both) or scalar path, in-LDS Cholesky,
//                                 back-substitution of the eliminated
//                                 feature.
//
// Stress-shape set (M > 64 or background > 128; round 2):
//   K2c pack_masks_words        — wide-M packing, one thread per output
//                                 word (coalesced 64B source spans)
//   K3-K6 fused_predict_tiled / fused_predict_tiled_bf16 —
//                                 column-tiled fused predict: diff streamed
//                                 through double-buffered LDS k-chunks
//                                 (software-pipelined: next chunk's global
//                                 loads issue during this chunk's MFMAs),
//                                 per-column-tile partial reductions summed
//                                 by a deterministic reduce kernel; bf16
//                                 variant runs one v_mfma_f32_16x16x32_bf16
//                                 per 32-deep k block with a hi+lo split B
//   K7b wls_gram                — (M-1) x (M-1+n_out) normal equations in
//                                 16x16 MFMA tiles over packed bits, fp64
//                                 via 256-sample chunked promotion; also
//                                 powers the batched L1 path (phantom last
//                                 feature) — the solve itself is library
//                                 fp64 (S-independent)
//
// All kernels are wave64 / LDS-tiled for CDNA4; fp32 compute by default,
// bf16 matrix-core modes opt-in (engine KernelConfig.predict_dtype).

#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64

// ------------------------------------------------------------------------- //
// Philox4x32-10 counter-based RNG (deterministic per (seed, instance, draw))
// ------------------------------------------------------------------------- //

struct Philox {
    uint32_t ctr[4];
    uint32_t key[2];
    uint32_t buf[4];
    int idx;

    __device__ void init(uint32_t k0, uint32_t k1, uint32_t c0, uint32_t c1) {
        key[0] = k0; key[1] = k1;
        ctr[0] = c0; ctr[1] = c1; ctr[2] = 0; ctr[3] = 0;
        idx = 4;
    }
    __device__ static void round_(uint32_t c[4], const uint32_t k[2]) {
        const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
        uint64_t p0 = (uint64_t)M0 * c[0];
        uint64_t p1 = (uint64_t)M1 * c[2];
        uint32_t h0 = (uint32_t)(p0 >> 32), l0 = (uint32_t)p0;
        uint32_t h1 = (uint32_t)(p1 >> 32), l1 = (uint32_t)p1;
        uint32_t n0 = h1 ^ c[1] ^ k[0];
        uint32_t n1 = l1;
        uint32_t n2 = h0 ^ c[3] ^ k[1];
        uint32_t n3 = l0;
        c[0] = n0; c[1] = n1; c[2] = n2; c[3] = n3;
    }
    __device__ void gen() {
        uint32_t c[4] = {ctr[0], ctr[1], ctr[2], ctr[3]};
        uint32_t k[2] = {key[0], key[1]};
        const uint32_t B0 = 0x9E3779B9u, B1 = 0xBB67AE85u;
#pragma unroll
        for (int r = 0; r < 10; ++r) {
            round_(c, k);
            k[0] += B0; k[1] += B1;
        }
        buf[0] = c[0]; buf[1] = c[1]; buf[2] = c[2]; buf[3] = c[3];
        ctr[2]++;                     // advance the stream
        if (ctr[2] == 0) ctr[3]++;
        idx = 0;
    }
    __device__ uint32_t next_u32() {
        if (idx >= 4) gen();
        return buf[idx++];
    }
    // uniform integer in [0, n) via mul-shift
    __device__ uint32_t next_below(uint32_t n) {
        return (uint32_t)(((uint64_t)next_u32() * n) >> 32);
    }
};

// ------------------------------------------------------------------------- //
// K2: random coalition masks.  One wave per instance; draws are complement-
// paired; the wave prefix-scans per-draw row consumption (1 or 2 rows) so all
// 64 lanes emit rows in parallel while preserving the sequential pair layout.
// ------------------------------------------------------------------------- //

template <int WORDS>   // bitset words: M <= 64*WORDS (WORDS <= 4)
__global__ void fill_random_masks_kernel(
    uint8_t* __restrict__ masks,      // (B, S, M)
    int B, int S, int M,
    int ne,                           // enumerated rows (already filled)
    int n_random,                     // rows to fill: [ne, ne+n_random)
    const float* __restrict__ cdf,    // (n_sizes,) cumulative probs
    const int* __restrict__ sizes,    // (n_sizes,) subset sizes
    int n_sizes,
    int num_paired,                   // sizes <= num_paired get a complement
    uint32_t seed,
    const int32_t* __restrict__ inst_ids,  // (B,) global instance index (RNG key)
    int n_chunks)                     // deterministic chunk count (multiple of 8)
{
    const int nblk = n_chunks / 8;    // blocks per instance, 8 waves each
    const int b = blockIdx.x / nblk;
    const int cb = blockIdx.x % nblk;
    if (b >= B) return;
    const int lane = threadIdx.x & (WAVE - 1);
    const int chunk_id = cb * 8 + (threadIdx.x >> 6);
    // fixed chunking (a deterministic function of n_random only) keeps the
    // output reproducible regardless of launch timing; complement pairs
    // never cross a chunk boundary.  n_chunks scales with n_random so big
    // plans (stress: 16k rows) fill the chip instead of B blocks
    const int chunk = (n_random + n_chunks - 1) / n_chunks;
    const int lo = chunk_id * chunk;
    const int hi = min(lo + chunk, n_random);
    if (lo >= hi) return;

    // size table read via L2/L3 (up to ceil((M-1)/2) <= 128 entries; a
    // fixed-size local copy silently truncates for wide M)
    uint8_t* mrow_base = masks + (size_t)b * S * M;

    int remaining = hi - lo;
    int written = lo;
    uint32_t iter = 0;
    while (remaining > 0) {
        Philox rng;
        rng.init(seed, (uint32_t)inst_ids[b], iter,
                 (uint32_t)(chunk_id * WAVE + lane) | 0x52000000u);
        // draw subset size from the residual kernel distribution
        float u = (rng.next_u32() >> 8) * (1.0f / 16777216.0f);
        int si = 0;
        while (si < n_sizes - 1 && u > cdf[si]) ++si;
        int ssize = sizes[si];
        bool paired = ssize <= num_paired;

        uint64_t pairmask = __ballot(paired);
        uint64_t below = pairmask & ((1ull << lane) - 1ull);
        int rows_before = lane + __popcll(below);
        int total_rows = WAVE + __popcll(pairmask);

        if (rows_before < remaining) {
            // sample ssize distinct bits by rejection (ssize <= ceil((M-1)/2)
            // so acceptance >= 1/2 per try)
            uint64_t bits[WORDS];
#pragma unroll
            for (int w = 0; w < WORDS; ++w) bits[w] = 0ull;
            int got = 0;
            int guard = 0;
            while (got < ssize && guard < 65536) {
                uint32_t r = rng.next_below((uint32_t)M);
                uint64_t bit = 1ull << (r & 63);
                if (!(bits[r >> 6] & bit)) { bits[r >> 6] |= bit; ++got; }
                ++guard;
            }
            uint8_t* row = mrow_base + (size_t)(ne + written + rows_before) * M;
            const bool wide = (M & 3) == 0;   // rows 4-aligned: dword stores
            if (wide) {
                for (int g = 0; g < M; g += 4) {
                    const uint32_t nib =
                        (uint32_t)((bits[g >> 6] >> (g & 63)) & 0xFull);
                    // bit i of the nibble -> byte i (mask bytes are 0/1)
                    *(uint32_t*)(row + g) =
                        (nib & 1u) | ((nib & 2u) << 7) | ((nib & 4u) << 14)
                        | ((nib & 8u) << 21);
                }
            } else {
                for (int g = 0; g < M; ++g)
                    row[g] = (uint8_t)((bits[g >> 6] >> (g & 63)) & 1ull);
            }
            if (paired && rows_before + 1 < remaining) {
                uint8_t* crow = row + M;
                if (wide) {
                    for (int g = 0; g < M; g += 4) {
                        const uint32_t nib = (uint32_t)(
                            (~bits[g >> 6] >> (g & 63)) & 0xFull);
                        *(uint32_t*)(crow + g) =
                            (nib & 1u) | ((nib & 2u) << 7) | ((nib & 4u) << 14)
                            | ((nib & 8u) << 21);
                    }
                } else {
                    for (int g = 0; g < M; ++g)
                        crow[g] = (uint8_t)(
                            1u - ((bits[g >> 6] >> (g & 63)) & 1ull));
                }
            }
        }
        int consumed = total_rows < remaining ? total_rows : remaining;
        written += consumed;
        remaining -= consumed;
        ++iter;
    }
}

extern "C" int kshap_sampler_chunks(int n_random)
{
    // deterministic chunk count: ~256 rows per chunk, multiple of 8 in
    // [8, 64] (n_random <= 2048 keeps the historic 8 chunks bit-for-bit)
    int c = (n_random + 255) / 256;
    c = (c + 7) / 8 * 8;
    if (c < 8) c = 8;
    if (c > 64) c = 64;
    return c;
}

extern "C" void launch_fill_random_masks(
    uint8_t* masks, int B, int S, int M, int ne, int n_random,
    const float* cdf, const int* sizes, int n_sizes, int num_paired,
    uint32_t seed, const int32_t* inst_ids, hipStream_t stream)
{
    if (n_random <= 0 || B <= 0) return;
    const int words = (M + 63) / 64;
    const int n_chunks = kshap_sampler_chunks(n_random);
    dim3 grid(B * (n_chunks / 8)), block(8 * WAVE);
    switch (words) {
        case 1:
            fill_random_masks_kernel<1><<<grid, block, 0, stream>>>(
                masks, B, S, M, ne, n_random, cdf, sizes, n_sizes, num_paired,
                seed, inst_ids, n_chunks);
            break;
        case 2:
            fill_random_masks_kernel<2><<<grid, block, 0, stream>>>(
                masks, B, S, M, ne, n_random, cdf, sizes, n_sizes, num_paired,
                seed, inst_ids, n_chunks);
            break;
        case 3:
            fill_random_masks_kernel<3><<<grid, block, 0, stream>>>(
                masks, B, S, M, ne, n_random, cdf, sizes, n_sizes, num_paired,
                seed, inst_ids, n_chunks);
            break;
        default:
            fill_random_masks_kernel<4><<<grid, block, 0, stream>>>(
                masks, B, S, M, ne, n_random, cdf, sizes, n_sizes, num_paired,
                seed, inst_ids, n_chunks);
            break;
    }
}

// ------------------------------------------------------------------------- //
// K3-K6 fused: ey[b,s,o] = sum_n wbg[n] * act( base[o,n]
//                               + sum_k mask[b,s,k] * diff[b,o,k,n] )
//
// MFMA mask@diff GEMM (v_mfma_f32_16x16x4_f32, A = 16 s-rows x 4 k,
// B = 4 k x 16 n-cols, C 16x16 fp32) with activation + weighted background
// reduction fused in the epilogue.  LDS row strides are chosen ≡16 mod 32
// words so the two k-rows read by a 32-lane group land on disjoint bank
// halves (conflict-free ds_read_b32).
//
// Template params: NOUT in {1,2,4}; ACT 0=none 1=sigmoid 2=softmax.
// ------------------------------------------------------------------------- //


// 1-ulp hardware reciprocal (v_rcp_f32): the epilogue sigmoids do not need
// the IEEE-correct 5-instruction division sequence hipcc emits for `/`
__device__ __forceinline__ float fast_rcp(float x) {
    return __builtin_amdgcn_rcpf(x);
}

typedef __attribute__((ext_vector_type(4))) float f32x4;

#define MAX_MPAD 64        // fused path supports up to 64 varying groups
#define S_TILE 512         // s-rows per workgroup (8 sub-tiles of 64)
#define S_SUB 64           // rows per sub-tile (4 waves x 16)

template <int NOUT, int ACT, int NT>   // NT = Npad/16 col tiles (constexpr
__global__ __launch_bounds__(256)      // LDS offsets: the kernel was VALU-
void fused_predict_linear_kernel(      // bound on runtime address math)
    const uint8_t* __restrict__ masksU, // (B, S, M) raw coalition masks
    const float* __restrict__ diff,     // (B, OIMG, Mpad, NT*16)
    const float* __restrict__ base,     // (OIMG, NT*16)
    const float* __restrict__ wbg,      // (NT*16)  0 for padding cols
    float* __restrict__ ey,             // (B, S, NOUT)
    int B, int S, int M, int Mpad)
{
    // ACT==3: binary softmax from the logit DIFFERENCE — one operand image
    // (z1 - z0), half the MFMAs, both outputs from sigma(z) in the epilogue
    constexpr int OIMG = (ACT == 3) ? 1 : NOUT;
    constexpr int NPAD = NT * 16;
    constexpr int NSTRIDE = NPAD + ((16 - (NPAD & 31)) & 31);  // ≡16 mod 32
    const int n_stiles = (S + S_TILE - 1) / S_TILE;
    const int b = blockIdx.x / n_stiles;
    const int stile = blockIdx.x % n_stiles;
    const int s0 = stile * S_TILE;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wave = tid >> 6;           // 0..3

    extern __shared__ float lds[];
    float* diff_lds = lds;                                   // OIMG*Mpad*NSTRIDE
    float* base_lds = diff_lds + OIMG * Mpad * NSTRIDE;      // OIMG*NPAD
    float* wbg_lds = base_lds + OIMG * NPAD;                 // NPAD

    // ---- stage diff / base / wbg once; the 8 s-subtiles reuse them --------
    const float* dsrc = diff + (size_t)b * OIMG * Mpad * NPAD;
    for (int idx = tid; idx < OIMG * Mpad * NPAD; idx += 256) {
        int ok = idx / NPAD;             // o * Mpad + k
        int n = idx % NPAD;
        diff_lds[ok * NSTRIDE + n] = dsrc[idx];
    }
    for (int idx = tid; idx < OIMG * NPAD; idx += 256) base_lds[idx] = base[idx];
    for (int idx = tid; idx < NPAD; idx += 256) wbg_lds[idx] = wbg[idx];
    __syncthreads();

    const int swave = wave * 16;         // this wave's 16 s-rows
    const int arow = lane & 15;          // A row (s) / B col (n) within tile
    const int akcol = lane >> 4;         // k within the 4-wide micro-step
    // per-lane LDS base pointer: all loop offsets are compile-time now
    const float* dbase = diff_lds + akcol * NSTRIDE + arow;
    // A-operand converted in-register straight from the raw u8 masks (every
    // element is consumed exactly once per launch; a dedicated f32 transposed
    // image cost a transpose kernel + 0.5 GB/step of HBM traffic)
    const uint8_t* mlane = masksU + ((size_t)b * S + swave + arow) * M;

    for (int sub = 0; sub < S_TILE / S_SUB; ++sub) {
        const int ssub0 = s0 + sub * S_SUB;
        if (ssub0 >= S) break;
        const int srow = ssub0 + swave + arow;
        const bool svalid = srow < S;

        // column tiles processed in two halves: halves re-read the (cheap,
        // L2-resident) A stream but halve the accumulator AGPR footprint,
        // buying occupancy (93 VGPR + 56 AGPR -> occ 3 at NT=7 otherwise)
        constexpr int NTH = (NT + 1) / 2;
        // ACT 3 accumulates only p1; ey0 = 1 - ey1 at the write (weights sum
        // to 1 over the real background columns)
        constexpr int NACC = (ACT == 3) ? 1 : NOUT;
        float partial[NACC][4];
#pragma unroll
        for (int o = 0; o < NACC; ++o)
#pragma unroll
            for (int r = 0; r < 4; ++r) partial[o][r] = 0.0f;

#pragma unroll
        for (int half = 0; half < 2; ++half) {
        const int CT0 = half * NTH;
        const int CTN = half == 0 ? NTH : NT - NTH;
        if (CTN <= 0) continue;

        f32x4 acc[NTH][OIMG];
#pragma unroll
        for (int ct = 0; ct < NTH; ++ct)
#pragma unroll
            for (int o = 0; o < OIMG; ++o) acc[ct][o] = (f32x4){0, 0, 0, 0};

        const uint8_t* mrow = mlane + (size_t)ssub0 * M;
        for (int ks = 0; ks < Mpad; ks += 4) {
            const int k = ks + akcol;
            float a = (svalid && k < M) ? (float)(mrow[k] & 1) : 0.0f;
#pragma unroll
            for (int ct = 0; ct < NTH; ++ct) {
                if (ct < CTN) {
#pragma unroll
                    for (int o = 0; o < OIMG; ++o) {
                        float bv = dbase[(o * Mpad + ks) * NSTRIDE + (CT0 + ct) * 16];
                        acc[ct][o] =
                            __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc[ct][o], 0, 0, 0);
                    }
                }
            }
        }

        // ---- epilogue: activation + weighted reduction over n -------------
        // C/D map: col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
        for (int ct = 0; ct < NTH; ++ct) {
            if (ct >= CTN) break;
            const int n = (CT0 + ct) * 16 + arow;
            float wn = wbg_lds[n];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float z[OIMG];
#pragma unroll
                for (int o = 0; o < OIMG; ++o) z[o] = acc[ct][o][r] + base_lds[o * NPAD + n];
                float zz[NACC];
                if (ACT == 3) {
                    zz[0] = fast_rcp(1.0f + __expf(-z[0]));   // p1 only
                } else if (ACT == 1) {
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) zz[o] = fast_rcp(1.0f + __expf(-z[o]));
                } else if (ACT == 2 && NOUT == 2) {
                    // binary softmax = one sigmoid: p1 = 1/(1+exp(z0-z1))
                    float p1 = fast_rcp(1.0f + __expf(z[0] - z[1]));
                    zz[0] = 1.0f - p1;
                    zz[1] = p1;
                } else if (ACT == 2) {
                    float mx = z[0];
#pragma unroll
                    for (int o = 1; o < NOUT; ++o) mx = fmaxf(mx, z[o]);
                    float sum = 0.0f;
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) { zz[o] = __expf(z[o] - mx); sum += zz[o]; }
                    float inv = fast_rcp(sum);
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) zz[o] *= inv;
                } else {
#pragma unroll
                    for (int o = 0; o < NACC; ++o) zz[o] = z[o];
                }
#pragma unroll
                for (int o = 0; o < NACC; ++o) partial[o][r] += wn * zz[o];
            }
        }
        }  // half loop
        // reduce over the 16 lanes of each row group (xor bits 0-3 in-group)
#pragma unroll
        for (int o = 0; o < NACC; ++o)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float v = partial[o][r];
                v += __shfl_xor(v, 1);
                v += __shfl_xor(v, 2);
                v += __shfl_xor(v, 4);
                v += __shfl_xor(v, 8);
                partial[o][r] = v;
            }
        if (arow == 0) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int ss = ssub0 + swave + akcol * 4 + r;
                if (ss < S) {
                    if (ACT == 3) {
                        float p1 = partial[0][r];
                        ey[((size_t)b * S + ss) * NOUT + 0] = 1.0f - p1;
                        ey[((size_t)b * S + ss) * NOUT + NOUT - 1] = p1;
                    } else {
#pragma unroll
                        for (int o = 0; o < NACC; ++o)
                            ey[((size_t)b * S + ss) * NOUT + o] = partial[o][r];
                    }
                }
            }
        }
    }  // sub-tile loop
}

template <int NOUT, int ACT>
static void launch_fused_nt(
    const uint8_t* masksU, const float* diff, const float* base, const float* wbg,
    float* ey, int B, int S, int M, int Mpad, int Npad, hipStream_t stream)
{
    const int n_stiles = (S + S_TILE - 1) / S_TILE;
    dim3 grid(B * n_stiles), block(256);
    const int NSTRIDE = Npad + ((16 - (Npad & 31)) & 31);
    const int oimg = (ACT == 3) ? 1 : NOUT;
    size_t lds = (size_t)(oimg * Mpad * NSTRIDE + oimg * Npad + Npad) * 4;
#define KSHAP_CASE(NTV) \
    case NTV: \
        fused_predict_linear_kernel<NOUT, ACT, NTV><<<grid, block, lds, stream>>>( \
            masksU, diff, base, wbg, ey, B, S, M, Mpad); \
        break;
    switch (Npad / 16) {
        KSHAP_CASE(1) KSHAP_CASE(2) KSHAP_CASE(3) KSHAP_CASE(4)
        KSHAP_CASE(5) KSHAP_CASE(6) KSHAP_CASE(7) KSHAP_CASE(8)
    }
#undef KSHAP_CASE
}

template <int NOUT>
static void launch_fused_act(
    const uint8_t* masksU, const float* diff, const float* base, const float* wbg,
    float* ey, int B, int S, int M, int Mpad, int Npad, int act, hipStream_t stream)
{
    switch (act) {
        case 0:
            launch_fused_nt<NOUT, 0>(masksU, diff, base, wbg, ey, B, S, M, Mpad, Npad, stream);
            break;
        case 1:
            launch_fused_nt<NOUT, 1>(masksU, diff, base, wbg, ey, B, S, M, Mpad, Npad, stream);
            break;
        case 3:
            if constexpr (NOUT == 2)
                launch_fused_nt<NOUT, 3>(masksU, diff, base, wbg, ey, B, S, M, Mpad, Npad, stream);
            break;
        default:
            launch_fused_nt<NOUT, 2>(masksU, diff, base, wbg, ey, B, S, M, Mpad, Npad, stream);
            break;
    }
}

extern "C" int launch_fused_predict_linear(
    const uint8_t* masksU, const float* diff, const float* base, const float* wbg,
    float* ey, int B, int S, int M, int Mpad, int Npad, int n_out, int act,
    hipStream_t stream)
{
    if (Mpad > MAX_MPAD || Npad % 16 != 0 || Npad / 16 > 8) return -1;
    switch (n_out) {
        case 1: launch_fused_act<1>(masksU, diff, base, wbg, ey, B, S, M, Mpad, Npad, act, stream); break;
        case 2: launch_fused_act<2>(masksU, diff, base, wbg, ey, B, S, M, Mpad, Npad, act, stream); break;
        case 4: launch_fused_act<4>(masksU, diff, base, wbg, ey, B, S, M, Mpad, Npad, act, stream); break;
        default: return -1;
    }
    return 0;
}

// ------------------------------------------------------------------------- //
// K3-K6 fused, bf16 matrix cores: one v_mfma_f32_16x16x32_bf16 covers the
// whole K dimension (Mpad <= 32) per column tile.  The A operand (coalition
// masks) is EXACT in bf16 (0/1); the B operand is either a single bf16
// image (predict_dtype='bf16') or a hi+lo split pair (predict_dtype=
// 'bf16x2': B = hi + (B - hi), two MFMAs, error ~2^-16 — fp32-grade).
// LDS B layout is k-major [o][n][KSTRIDE] so each lane's 8-element k-block
// is one ds_read_b128; KSTRIDE=40 staggers the 16-lane groups across banks.
// ------------------------------------------------------------------------- //

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define KSTRIDE_BF 40   // 32 k slots + 8 pad: (20*n)%64 distinct for n=0..15

__global__ void pack_masks_kernel(
    const uint8_t* __restrict__ masks,  // (B, S, M)
    uint64_t* __restrict__ packed,      // (B, S)
    size_t n_rows, int M)
{
    // one thread per row, bytes gathered with dword loads (a thread-per-bit
    // ballot variant was load-issue-bound: 32 byte loads per 12-byte row)
    const size_t row = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (row >= n_rows) return;
    const uint8_t* src = masks + row * M;
    uint64_t bits = 0ull;
    int k = 0;
    // rows are 4-byte aligned whenever M % 4 == 0 (the common case)
    if (((size_t)src & 3) == 0) {
        for (; k + 4 <= M; k += 4) {
            uint32_t w = *(const uint32_t*)(src + k);
#pragma unroll
            for (int j = 0; j < 4; ++j)
                bits |= ((uint64_t)((w >> (8 * j)) & 1u)) << (k + j);
        }
    }
    for (; k < M; ++k) bits |= ((uint64_t)(src[k] & 1)) << k;
    packed[row] = bits;
}

extern "C" void launch_pack_masks(
    const uint8_t* masks, uint64_t* packed, int B, int S, int M,
    hipStream_t stream)
{
    size_t n = (size_t)B * S;
    dim3 grid((unsigned)((n + 255) / 256)), block(256);
    pack_masks_kernel<<<grid, block, 0, stream>>>(masks, packed, n, M);
}

template <int NOUT, int ACT, int NT, int SPLIT>  // SPLIT: 1 = hi only, 2 = hi+lo
__global__ __launch_bounds__(256)
void fused_predict_bf16_kernel(
    const uint8_t* __restrict__ masksU, // (B, S, M) raw coalition masks
    const __bf16* __restrict__ diffB,   // (B, SPLIT, NOUT, NT*16, KSTRIDE_BF)
    const float* __restrict__ base,     // (NOUT, NT*16)
    const float* __restrict__ wbg,      // (NT*16)
    float* __restrict__ ey,             // (B, S, NOUT)
    int B, int S, int M)
{
    constexpr int OIMG = (ACT == 3) ? 1 : NOUT;   // ACT 3: logit-difference
    constexpr int NPAD = NT * 16;
    const int n_stiles = (S + S_TILE - 1) / S_TILE;
    const int b = blockIdx.x / n_stiles;
    const int stile = blockIdx.x % n_stiles;
    const int s0 = stile * S_TILE;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wave = tid >> 6;

    extern __shared__ float lds[];
    __bf16* diff_lds = (__bf16*)lds;                 // SPLIT*OIMG*NPAD*KSTRIDE_BF
    float* base_lds = lds + (SPLIT * OIMG * NPAD * KSTRIDE_BF + 1) / 2;
    float* wbg_lds = base_lds + OIMG * NPAD;

    const __bf16* dsrc = diffB + (size_t)b * SPLIT * OIMG * NPAD * KSTRIDE_BF;
    for (int idx = tid; idx < SPLIT * OIMG * NPAD * KSTRIDE_BF / 8; idx += 256)
        ((bf16x8*)diff_lds)[idx] = ((const bf16x8*)dsrc)[idx];
    for (int idx = tid; idx < OIMG * NPAD; idx += 256) base_lds[idx] = base[idx];
    for (int idx = tid; idx < NPAD; idx += 256) wbg_lds[idx] = wbg[idx];
    __syncthreads();

    const int swave = wave * 16;
    const int arow = lane & 15;          // A row (s) / B col (n)
    const int akb = lane >> 4;           // k-block 0..3 (8 elements each)
    const uint8_t* mlane = masksU + ((size_t)b * S + swave + arow) * M;
    // per-lane LDS base for B fragments: [split][o][n=ct*16+arow][k=akb*8]
    const __bf16* dlane = diff_lds + (size_t)arow * KSTRIDE_BF + akb * 8;

    for (int sub = 0; sub < S_TILE / S_SUB; ++sub) {
        const int ssub0 = s0 + sub * S_SUB;
        if (ssub0 >= S) break;
        const int srow = ssub0 + swave + arow;
        const bool svalid = srow < S;
        // A fragment converted in-register from the raw u8 masks (exact in
        // bf16); a dedicated bf16 mask image would cost 1.4 GB/step of
        // HBM traffic at B=10k
        const uint8_t* mrow = mlane + (size_t)ssub0 * M;
        bf16x8 a;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int k = akb * 8 + j;
            a[j] = (__bf16)(float)(
                (svalid && k < M) ? (mrow[k] & 1) : 0);
        }

        constexpr int NACC = (ACT == 3) ? 1 : NOUT;
        float partial[NACC][4];
#pragma unroll
        for (int o = 0; o < NACC; ++o)
#pragma unroll
            for (int r = 0; r < 4; ++r) partial[o][r] = 0.0f;

#pragma unroll
        for (int ct = 0; ct < NT; ++ct) {
            f32x4 acc[OIMG];
#pragma unroll
            for (int o = 0; o < OIMG; ++o) {
                acc[o] = (f32x4){0, 0, 0, 0};
#pragma unroll
                for (int sp = 0; sp < SPLIT; ++sp) {
                    bf16x8 bv = *(const bf16x8*)(
                        dlane + ((size_t)(sp * OIMG + o) * NPAD + ct * 16) * KSTRIDE_BF);
                    acc[o] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a, bv, acc[o], 0, 0, 0);
                }
            }
            const int n = ct * 16 + arow;
            float wn = wbg_lds[n];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float z[OIMG];
#pragma unroll
                for (int o = 0; o < OIMG; ++o) z[o] = acc[o][r] + base_lds[o * NPAD + n];
                float zz[NACC];
                if (ACT == 3) {
                    zz[0] = fast_rcp(1.0f + __expf(-z[0]));   // p1 only
                } else if (ACT == 1) {
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) zz[o] = fast_rcp(1.0f + __expf(-z[o]));
                } else if (ACT == 2 && NOUT == 2) {
                    float p1 = fast_rcp(1.0f + __expf(z[0] - z[1]));
                    zz[0] = 1.0f - p1;
                    zz[1] = p1;
                } else if (ACT == 2) {
                    float mx = z[0];
#pragma unroll
                    for (int o = 1; o < NOUT; ++o) mx = fmaxf(mx, z[o]);
                    float sum = 0.0f;
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) { zz[o] = __expf(z[o] - mx); sum += zz[o]; }
                    float inv = fast_rcp(sum);
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) zz[o] *= inv;
                } else {
#pragma unroll
                    for (int o = 0; o < NACC; ++o) zz[o] = z[o];
                }
#pragma unroll
                for (int o = 0; o < NACC; ++o) partial[o][r] += wn * zz[o];
            }
        }
#pragma unroll
        for (int o = 0; o < NACC; ++o)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float v = partial[o][r];
                v += __shfl_xor(v, 1);
                v += __shfl_xor(v, 2);
                v += __shfl_xor(v, 4);
                v += __shfl_xor(v, 8);
                partial[o][r] = v;
            }
        if (arow == 0) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int ss = ssub0 + swave + akb * 4 + r;
                if (ss < S) {
                    if (ACT == 3) {
                        float p1 = partial[0][r];
                        ey[((size_t)b * S + ss) * NOUT + 0] = 1.0f - p1;
                        ey[((size_t)b * S + ss) * NOUT + NOUT - 1] = p1;
                    } else {
#pragma unroll
                        for (int o = 0; o < NACC; ++o)
                            ey[((size_t)b * S + ss) * NOUT + o] = partial[o][r];
                    }
                }
            }
        }
    }
}

template <int NOUT, int ACT>
static void launch_fused_bf16_nt(
    const uint8_t* masksU, const __bf16* diffB, const float* base,
    const float* wbg, float* ey, int B, int S, int M, int Npad, int split,
    hipStream_t stream)
{
    const int n_stiles = (S + S_TILE - 1) / S_TILE;
    dim3 grid(B * n_stiles), block(256);
    const int oimg = (ACT == 3) ? 1 : NOUT;
    size_t lds = ((size_t)split * oimg * Npad * KSTRIDE_BF * 2 + 2)
                 + (size_t)(oimg * Npad + Npad) * 4;
    lds = (lds + 3) / 4 * 4 + 4;
#define KSHAP_BF_CASE(NTV, SPL) \
    if (Npad / 16 == NTV && split == SPL) { \
        fused_predict_bf16_kernel<NOUT, ACT, NTV, SPL><<<grid, block, lds, stream>>>( \
            masksU, diffB, base, wbg, ey, B, S, M); \
        return; \
    }
    KSHAP_BF_CASE(1,1) KSHAP_BF_CASE(2,1) KSHAP_BF_CASE(3,1) KSHAP_BF_CASE(4,1)
    KSHAP_BF_CASE(5,1) KSHAP_BF_CASE(6,1) KSHAP_BF_CASE(7,1) KSHAP_BF_CASE(8,1)
    KSHAP_BF_CASE(1,2) KSHAP_BF_CASE(2,2) KSHAP_BF_CASE(3,2) KSHAP_BF_CASE(4,2)
    KSHAP_BF_CASE(5,2) KSHAP_BF_CASE(6,2) KSHAP_BF_CASE(7,2) KSHAP_BF_CASE(8,2)
#undef KSHAP_BF_CASE
}

template <int NOUT>
static void launch_fused_bf16_act(
    const uint8_t* masksU, const __bf16* diffB, const float* base,
    const float* wbg, float* ey, int B, int S, int M, int Npad, int act,
    int split, hipStream_t stream)
{
    switch (act) {
        case 0: launch_fused_bf16_nt<NOUT, 0>(masksU, diffB, base, wbg, ey, B, S, M, Npad, split, stream); break;
        case 1: launch_fused_bf16_nt<NOUT, 1>(masksU, diffB, base, wbg, ey, B, S, M, Npad, split, stream); break;
        case 3:
            if constexpr (NOUT == 2)
                launch_fused_bf16_nt<NOUT, 3>(masksU, diffB, base, wbg, ey, B, S, M, Npad, split, stream);
            break;
        default: launch_fused_bf16_nt<NOUT, 2>(masksU, diffB, base, wbg, ey, B, S, M, Npad, split, stream); break;
    }
}

extern "C" int launch_fused_predict_bf16(
    const uint8_t* masksU, const uint16_t* diffB_u, const float* base,
    const float* wbg, float* ey, int B, int S, int M, int Npad, int n_out,
    int act, int split, hipStream_t stream)
{
    if (M > 32 || Npad % 16 != 0 || Npad / 16 > 8 || split < 1 || split > 2)
        return -1;
    const __bf16* diffB = reinterpret_cast<const __bf16*>(diffB_u);
    switch (n_out) {
        case 1: launch_fused_bf16_act<1>(masksU, diffB, base, wbg, ey, B, S, M, Npad, act, split, stream); break;
        case 2: launch_fused_bf16_act<2>(masksU, diffB, base, wbg, ey, B, S, M, Npad, act, split, stream); break;
        case 4: launch_fused_bf16_act<4>(masksU, diffB, base, wbg, ey, B, S, M, Npad, act, split, stream); break;
        default: return -1;
    }
    return 0;
}

// ------------------------------------------------------------------------- //
// diff-image builders: diff[k, n] = x_part[b, vidx[k], o] - bg_part[n,
// vidx[k], o], written directly in the fused kernels\' operand layouts
// (replaces a chain of strided torch scatter kernels inside the graph).
// ------------------------------------------------------------------------- //

__global__ void build_diff_f32_kernel(
    const float* __restrict__ xp,    // (B, G, O)
    const float* __restrict__ bgp,   // (N, G, O)
    const int64_t* __restrict__ vidx,  // (m,)
    float* __restrict__ out,         // (B, O, Mpad, Npad) — pad slots stay 0
    int G, int O, int N, int m, int Mpad, int Npad, size_t total)
{
    size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;       // total = B*O*m*N, n fastest
    const int n = idx % N;
    const int k = (idx / N) % m;
    const int o = (idx / ((size_t)N * m)) % O;
    const size_t b = idx / ((size_t)N * m * O);
    const int64_t g = vidx[k];
    float v = xp[((size_t)b * G + g) * O + o] - bgp[((size_t)n * G + g) * O + o];
    out[(((size_t)b * O + o) * Mpad + k) * Npad + n] = v;
}

extern "C" void launch_build_diff_f32(
    const float* xp, const float* bgp, const int64_t* vidx, float* out,
    int B, int G, int O, int N, int m, int Mpad, int Npad, hipStream_t stream)
{
    size_t total = (size_t)B * O * m * N;
    build_diff_f32_kernel<<<dim3((unsigned)((total + 255) / 256)), dim3(256), 0,
                            stream>>>(xp, bgp, vidx, out, G, O, N, m, Mpad,
                                      Npad, total);
}

__global__ void build_diff_bf16_kernel(
    const float* __restrict__ xp,    // (B, G, O)
    const float* __restrict__ bgp,   // (N, G, O)
    const int64_t* __restrict__ vidx,  // (m,)
    __bf16* __restrict__ out,        // (B, SPLIT, O, Npad, KSTRIDE_BF)
    int G, int O, int N, int m, int Npad, int split, size_t total)
{
    size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;       // total = B*O*N*m, k fastest
    const int k = idx % m;
    const int n = (idx / m) % N;
    const int o = (idx / ((size_t)m * N)) % O;
    const size_t b = idx / ((size_t)m * N * O);
    const int64_t g = vidx[k];
    float v = xp[((size_t)b * G + g) * O + o] - bgp[((size_t)n * G + g) * O + o];
    __bf16 hi = (__bf16)v;
    size_t base = ((((size_t)b * split) * O + o) * Npad + n) * KSTRIDE_BF + k;
    out[base] = hi;
    if (split == 2) {
        size_t lobase = ((((size_t)b * split + 1) * O + o) * Npad + n) * KSTRIDE_BF + k;
        out[lobase] = (__bf16)(v - (float)hi);
    }
}

extern "C" void launch_build_diff_bf16(
    const float* xp, const float* bgp, const int64_t* vidx, uint16_t* out,
    int B, int G, int O, int N, int m, int Npad, int split, hipStream_t stream)
{
    size_t total = (size_t)B * O * N * m;
    build_diff_bf16_kernel<<<dim3((unsigned)((total + 255) / 256)), dim3(256),
                             0, stream>>>(
        xp, bgp, vidx, reinterpret_cast<__bf16*>(out), G, O, N, m, Npad,
        split, total);
}

// ------------------------------------------------------------------------- //
// K3': explicit masked-background synthesis for the torch-predictor path.
// out[(s - s_lo)*N + n, d] = mask[b, s, group(d)] ? x[b, d] : bg[n, d]
// Coalesced over d; one workgroup covers one (s, n) row-pair block.
// ------------------------------------------------------------------------- //

template <typename T>
__global__ void synth_chunk_kernel(
    const uint8_t* __restrict__ masks,   // (B, S, M)
    const float* __restrict__ x,         // (B, D)
    const float* __restrict__ bg,        // (N, D)
    const int* __restrict__ col_group,   // (D)
    T* __restrict__ out,                 // ((b_hi-b_lo)*(s_hi-s_lo)*N, D)
    int S, int M, int N, int D, int b_lo, int b_hi, int s_lo, int s_hi)
{
    // one WAVE per row, grid-strided (a block-per-row launch at the mlp
    // config dispatched 55.7M 64-thread blocks per step and ran at 1.7 TB/s)
    const int srange = s_hi - s_lo;
    const size_t nrows = (size_t)(b_hi - b_lo) * srange * N;
    const int wv = threadIdx.x >> 6;     // 4 waves per block
    const int lane = threadIdx.x & 63;
    for (size_t row = (size_t)blockIdx.x * 4 + wv; row < nrows;
         row += (size_t)gridDim.x * 4) {
        const int n = row % N;
        const int s = s_lo + (row / N) % srange;
        const int b = b_lo + row / ((size_t)N * srange);
        const uint8_t* mrow = masks + ((size_t)b * S + s) * M;
        const float* xrow = x + (size_t)b * D;
        const float* brow = bg + (size_t)n * D;
        T* orow = out + (size_t)row * D;
        for (int d = lane; d < D; d += 64)
            orow[d] = (T)(mrow[col_group[d]] ? xrow[d] : brow[d]);
    }
}

extern "C" void launch_synth_chunk(
    const uint8_t* masks, const float* x, const float* bg, const int* col_group,
    void* out, int out_bf16, int S, int M, int N, int D, int b_lo, int b_hi,
    int s_lo, int s_hi, hipStream_t stream)
{
    size_t nrows = (size_t)(b_hi - b_lo) * (s_hi - s_lo) * N;
    // ~8 rows per wave; cap the grid so huge sweeps grid-stride
    size_t blocks = (nrows + 31) / 32;
    if (blocks > (1u << 20)) blocks = 1u << 20;
    if (blocks == 0) return;
    // bf16 output feeds autocast modules directly: halves the synth write
    // traffic AND removes the separate fp32->bf16 cast pass torch would run
    // over the whole perturbation tensor (mlp profile: 12.7 ms synth +
    // 18 ms elementwise per step)
    if (out_bf16)
        synth_chunk_kernel<__bf16><<<dim3((unsigned)blocks), dim3(256), 0,
                                     stream>>>(
            masks, x, bg, col_group, (__bf16*)out, S, M, N, D, b_lo, b_hi,
            s_lo, s_hi);
    else
        synth_chunk_kernel<float><<<dim3((unsigned)blocks), dim3(256), 0,
                                    stream>>>(
            masks, x, bg, col_group, (float*)out, S, M, N, D, b_lo, b_hi,
            s_lo, s_hi);
}

// ------------------------------------------------------------------------- //
// K7: batched constrained WLS solve.  One 256-thread workgroup per instance.
//   etmp[s,i] = mask[s,i] - mask[s,last]           (i < M-1)
//   ey2[s,o]  = eyAdj[s,o] - mask[s,last]*total[o]
//   A = etmp^T diag(w) etmp;  r_o = etmp^T (w * ey2_o)
//   A w = r (Cholesky);  phi[last] = total - sum(w)
// Masks are packed to uint64 bitfields in LDS per 256-sample chunk; each
// thread owns a strided set of Gram entries (upper triangle) / rhs entries.
// ------------------------------------------------------------------------- //

#define WLS_CHUNK 256
#define WLS_MAX_M 64
#define WLS_MAX_NOUT 8

__global__ __launch_bounds__(256)
void wls_solve_kernel(
    const uint8_t* __restrict__ masks,   // (B, S, M)
    const float* __restrict__ kw,        // (B, S)
    const float* __restrict__ ey_adj,    // (B, S, n_out)
    const float* __restrict__ total,     // (B, n_out)
    float* __restrict__ phi,             // (B, M, n_out)
    int B, int S, int M, int n_out)
{
    const int b = blockIdx.x;
    if (b >= B) return;
    const int tid = threadIdx.x;
    const int mm = M - 1;
    const int npairs = mm * (mm + 1) / 2;

    __shared__ uint64_t pk[WLS_CHUNK];
    __shared__ float wch[WLS_CHUNK];
    __shared__ float eych[WLS_CHUNK][WLS_MAX_NOUT];
    __shared__ float A[WLS_MAX_M * WLS_MAX_M];
    __shared__ float rhs[WLS_MAX_M * WLS_MAX_NOUT];
    __shared__ float tot_s[WLS_MAX_NOUT];

    if (tid < n_out) tot_s[tid] = total[(size_t)b * n_out + tid];
    __syncthreads();

    // per-thread accumulators over its strided entries
    float accA[8];        // up to 8 pairs per thread: npairs <= 2016, 256 thr
    float accR[8];
    const int pairs_per_thread = (npairs + 255) / 256;
    const int rtot = mm * n_out;
    const int r_per_thread = (rtot + 255) / 256;
    for (int q = 0; q < 8; ++q) { accA[q] = 0.0f; accR[q] = 0.0f; }

    const uint8_t* mbase = masks + (size_t)b * S * M;
    const float* kwb = kw + (size_t)b * S;
    const float* eyb = ey_adj + (size_t)b * S * n_out;

    for (int c0 = 0; c0 < S; c0 += WLS_CHUNK) {
        const int clen = min(WLS_CHUNK, S - c0);
        __syncthreads();
        if (tid < clen) {
            const uint8_t* mrow = mbase + (size_t)(c0 + tid) * M;
            uint64_t bits = 0ull;
            for (int g = 0; g < M; ++g) bits |= ((uint64_t)(mrow[g] & 1)) << g;
            pk[tid] = bits;
            wch[tid] = kwb[c0 + tid];
            float mlast = (float)((bits >> (M - 1)) & 1ull);
            for (int o = 0; o < n_out; ++o)
                eych[tid][o] = eyb[(size_t)(c0 + tid) * n_out + o] - mlast * tot_s[o];
        }
        __syncthreads();
        // Gram entries
        for (int q = 0; q < pairs_per_thread; ++q) {
            int p = tid + q * 256;
            if (p >= npairs) break;
            // unrank upper-triangle pair (i <= j)
            int i = 0, rem = p;
            while (rem >= mm - i) { rem -= mm - i; ++i; }
            int j = i + rem;
            float s_acc = 0.0f;
            for (int t = 0; t < clen; ++t) {
                uint64_t bits = pk[t];
                float ml = (float)((bits >> (M - 1)) & 1ull);
                float ei = (float)((bits >> i) & 1ull) - ml;
                float ej = (float)((bits >> j) & 1ull) - ml;
                s_acc += wch[t] * ei * ej;
            }
            accA[q] += s_acc;
        }
        // rhs entries
        for (int q = 0; q < r_per_thread; ++q) {
            int p = tid + q * 256;
            if (p >= rtot) break;
            int i = p / n_out, o = p % n_out;
            float s_acc = 0.0f;
            for (int t = 0; t < clen; ++t) {
                uint64_t bits = pk[t];
                float ml = (float)((bits >> (M - 1)) & 1ull);
                float ei = (float)((bits >> i) & 1ull) - ml;
                s_acc += wch[t] * ei * eych[t][o];
            }
            accR[q] += s_acc;
        }
    }
    __syncthreads();
    // scatter accumulators into LDS A (symmetric) and rhs
    for (int q = 0; q < pairs_per_thread; ++q) {
        int p = tid + q * 256;
        if (p >= npairs) break;
        int i = 0, rem = p;
        while (rem >= mm - i) { rem -= mm - i; ++i; }
        int j = i + rem;
        A[i * mm + j] = accA[q];
        A[j * mm + i] = accA[q];
    }
    for (int q = 0; q < r_per_thread; ++q) {
        int p = tid + q * 256;
        if (p >= rtot) break;
        rhs[p] = accR[q];
    }
    __syncthreads();

    // Cholesky factorisation (thread 0; mm <= 63, cold relative to Gram build)
    if (tid == 0) {
        for (int k = 0; k < mm; ++k) {
            float d = A[k * mm + k];
            for (int t = 0; t < k; ++t) d -= A[k * mm + t] * A[k * mm + t];
            d = sqrtf(fmaxf(d, 1e-20f));
            A[k * mm + k] = d;
            float inv = 1.0f / d;
            for (int r = k + 1; r < mm; ++r) {
                float v = A[r * mm + k];
                for (int t = 0; t < k; ++t) v -= A[r * mm + t] * A[k * mm + t];
                A[r * mm + k] = v * inv;
            }
        }
    }
    __syncthreads();
    // triangular solves: one thread per output
    if (tid < n_out) {
        const int o = tid;
        float y[WLS_MAX_M];
        for (int i = 0; i < mm; ++i) {
            float v = rhs[i * n_out + o];
            for (int t = 0; t < i; ++t) v -= A[i * mm + t] * y[t];
            y[i] = v / A[i * mm + i];
        }
        float w[WLS_MAX_M];
        float sumw = 0.0f;
        for (int i = mm - 1; i >= 0; --i) {
            float v = y[i];
            for (int t = i + 1; t < mm; ++t) v -= A[t * mm + i] * w[t];
            w[i] = v / A[i * mm + i];
        }
        for (int i = 0; i < mm; ++i) sumw += w[i];
        float* prow = phi + (size_t)b * M * n_out;
        for (int i = 0; i < mm; ++i) prow[i * n_out + o] = w[i];
        prow[(M - 1) * n_out + o] = tot_s[o] - sumw;
    }
}


// ------------------------------------------------------------------------- //
// K7-MFMA: Gram + rhs build on matrix cores for mm + n_out <= 16.
// One 16x16x4 f32 MFMA accumulates BOTH the (mm x mm) Gram matrix and the
// (mm x n_out) rhs per 4 samples:  A[i][k] = e[k][i],
// B[k][j] = w[k] * (j < mm ? e[k][j] : ey2[k][j-mm]).
// 4 waves split the sample (K) axis; partial tiles summed through LDS, then
// the same Cholesky/solve epilogue as the generic kernel.
// ------------------------------------------------------------------------- //

__global__ __launch_bounds__(256)
void wls_solve_mfma_kernel(
    const uint64_t* __restrict__ packed, // (B, S) pre-packed mask bits
    const float* __restrict__ kw,        // (B, S)
    const float* __restrict__ ey_adj,    // (B, S, n_out)
    const float* __restrict__ total,     // (B, n_out)
    float* __restrict__ phi,             // (B, M, n_out)
    int B, int S, int M, int n_out)
{
    const int b = blockIdx.x;
    if (b >= B) return;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wv = tid >> 6;             // wave 0..3
    const int mm = M - 1;
    const int cols = mm + n_out;         // <= 16

    __shared__ uint64_t pk[WLS_CHUNK];
    __shared__ float wch[WLS_CHUNK];
    __shared__ float eych[WLS_CHUNK][WLS_MAX_NOUT];
    __shared__ float tileA[4][16][17];   // per-wave 16x16 (+1 pad)
    __shared__ float A[16 * 16];
    __shared__ float rhs[16 * WLS_MAX_NOUT];
    __shared__ float tot_s[WLS_MAX_NOUT];

    if (tid < n_out) tot_s[tid] = total[(size_t)b * n_out + tid];
    __syncthreads();                     // tot_s visible to every wave

    const uint64_t* pbase = packed + (size_t)b * S;
    const float* kwb = kw + (size_t)b * S;
    const float* eyb = ey_adj + (size_t)b * S * n_out;

    const int arow = lane & 15;          // i (Gram row)
    const int akk = lane >> 4;           // k within the 4-sample micro-step

    f32x4 acc = (f32x4){0, 0, 0, 0};

    for (int c0 = 0; c0 < S; c0 += WLS_CHUNK) {
        const int clen = min(WLS_CHUNK, S - c0);
        // thread tid = wv*64 + lane stages exactly the sample its own wave
        // consumes below, so chunk staging needs NO barriers — an lgkmcnt
        // wait makes the wave's LDS writes visible to its own lanes
        if (tid < clen) {
            uint64_t bits = pbase[c0 + tid];
            pk[tid] = bits;
            wch[tid] = kwb[c0 + tid];
            float mlast = (float)((bits >> (M - 1)) & 1ull);
            for (int o = 0; o < n_out; ++o)
                eych[tid][o] = eyb[(size_t)(c0 + tid) * n_out + o] - mlast * tot_s[o];
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        // wave wv covers samples [wv*64, wv*64+64) of the chunk, 4 per step
        const int base = wv * 64;
        for (int ks = 0; ks < 64; ks += 4) {
            int t = base + ks + akk;     // chunk-local sample id
            float a = 0.0f, bv = 0.0f;
            if (t < clen) {
                uint64_t bits = pk[t];
                float ml = (float)((bits >> (M - 1)) & 1ull);
                float w = wch[t];
                if (arow < mm) a = (float)((bits >> arow) & 1ull) - ml;
                if (arow < mm) bv = w * ((float)((bits >> arow) & 1ull) - ml);
                else if (arow < cols) bv = w * eych[t][arow - mm];
            }
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
        }
    }
    // C/D map: col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
    for (int r = 0; r < 4; ++r)
        tileA[wv][(lane >> 4) * 4 + r][lane & 15] = acc[r];
    __syncthreads();
    for (int idx = tid; idx < 16 * 16; idx += 256) {
        int i = idx / 16, j = idx % 16;
        float v = tileA[0][i][j] + tileA[1][i][j] + tileA[2][i][j] + tileA[3][i][j];
        if (j < mm) A[i * mm + j] = v;           // Gram
        else if (j < cols && i < mm) rhs[i * n_out + (j - mm)] = v;  // rhs
    }
    __syncthreads();

    if (tid == 0) {
        for (int k = 0; k < mm; ++k) {
            float d = A[k * mm + k];
            for (int t = 0; t < k; ++t) d -= A[k * mm + t] * A[k * mm + t];
            d = sqrtf(fmaxf(d, 1e-20f));
            A[k * mm + k] = d;
            float inv = 1.0f / d;
            for (int r = k + 1; r < mm; ++r) {
                float v = A[r * mm + k];
                for (int t = 0; t < k; ++t) v -= A[r * mm + t] * A[k * mm + t];
                A[r * mm + k] = v * inv;
            }
        }
    }
    __syncthreads();
    if (tid < n_out) {
        const int o = tid;
        float y[16];
        for (int i = 0; i < mm; ++i) {
            float v = rhs[i * n_out + o];
            for (int t = 0; t < i; ++t) v -= A[i * mm + t] * y[t];
            y[i] = v / A[i * mm + i];
        }
        float w[16];
        float sumw = 0.0f;
        for (int i = mm - 1; i >= 0; --i) {
            float v = y[i];
            for (int t = i + 1; t < mm; ++t) v -= A[t * mm + i] * w[t];
            w[i] = v / A[i * mm + i];
        }
        for (int i = 0; i < mm; ++i) sumw += w[i];
        float* prow = phi + (size_t)b * M * n_out;
        for (int i = 0; i < mm; ++i) prow[i * n_out + o] = w[i];
        prow[(M - 1) * n_out + o] = tot_s[o] - sumw;
    }
}

extern "C" int launch_wls_solve(
    const uint8_t* masks, const uint64_t* packed, const float* kw,
    const float* ey_adj, const float* total, float* phi, int B, int S, int M,
    int n_out, hipStream_t stream)
{
    if (M < 2 || M > WLS_MAX_M || n_out > WLS_MAX_NOUT) return -1;
    int mm = M - 1;
    if (mm * (mm + 1) / 2 > 8 * 256) return -1;
    if (mm + n_out <= 16 && packed != nullptr) {
        // (a full-S LDS-staged variant was tried and measured SLOWER —
        // 222 vs 120 us at S=2072: the 47 KB footprint cut occupancy; the
        // 8-chunk 17 KB version wins on wave overlap)
        wls_solve_mfma_kernel<<<dim3(B), dim3(256), 0, stream>>>(
            packed, kw, ey_adj, total, phi, B, S, M, n_out);
    } else {
        wls_solve_kernel<<<dim3(B), dim3(256), 0, stream>>>(
            masks, kw, ey_adj, total, phi, B, S, M, n_out);
    }
    return 0;
}

// ------------------------------------------------------------------------- //
// K2b (wide): u8 mask rows -> W-word packed bitfields, M up to 64*W.
// Feeds the tiled MFMA Gram build for the stress shapes (M up to 513).
// ------------------------------------------------------------------------- //

__global__ void pack_masks_words_kernel(
    const uint8_t* __restrict__ masks,  // (B, S, M)
    uint64_t* __restrict__ packed,      // (B, S, W)
    size_t n_items, int M, int W)
{
    // one thread per OUTPUT WORD (row, w): each thread reads its 64-byte
    // source span — exactly one cache line when M%64==0 — and consecutive
    // threads (w fastest) touch consecutive lines, so both the mask reads
    // and the packed writes coalesce (the thread-per-row variant issued
    // 256-byte-strided dword loads: 16x line amplification at M=256)
    const size_t item = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (item >= n_items) return;
    const size_t row = item / W;
    const int w = (int)(item % W);
    const uint8_t* src = masks + row * M;
    const int k0 = w * 64;
    const int k1 = min(M, k0 + 64);
    uint64_t bits = 0ull;
    int k = k0;
    if ((((size_t)(src + k0)) & 3) == 0) {
        for (; k + 4 <= k1; k += 4) {
            uint32_t word = *(const uint32_t*)(src + k);
#pragma unroll
            for (int j = 0; j < 4; ++j)
                bits |= ((uint64_t)((word >> (8 * j)) & 1u)) << (k - k0 + j);
        }
    }
    for (; k < k1; ++k) bits |= ((uint64_t)(src[k] & 1)) << (k - k0);
    packed[row * W + w] = bits;
}

extern "C" void launch_pack_masks_words(
    const uint8_t* masks, uint64_t* packed, int B, int S, int M, int W,
    hipStream_t stream)
{
    size_t n = (size_t)B * S * W;
    pack_masks_words_kernel<<<dim3((unsigned)((n + 255) / 256)), dim3(256), 0,
                              stream>>>(masks, packed, n, M, W);
}

// ------------------------------------------------------------------------- //
// K7 (stress shapes): tiled MFMA Gram + rhs build for mm = M-1 up to 512.
//
//   A[i][j]   = sum_s w[s] * e[s][i] * e[s][j]        (i, j < mm)
//   rhs[i][o] = sum_s w[s] * e[s][i] * ey2[s][o]      (cols j = mm..mm+n_out)
//   with e[s][i] = mask[s][i] - mask[s][last], ey2 = eyAdj - mask_last*total
//
// One workgroup per (instance, 16-row tile strip).  The strip loops 16-col
// tile blocks (4 at a time); per block it scans S in 256-sample chunks whose
// packed bits/weights/ey2 are staged in LDS, accumulating each 16x16 tile
// with v_mfma_f32_16x16x4_f32 and PROMOTING the chunk partial to fp64
// registers — the in-fp32 accumulation length is <= 256, so the Gram reaches
// the fp64 torch Cholesky/solve at fp64-grade accuracy despite the Shapley
// kernel weights spanning orders of magnitude.  Only
// upper-triangle tile blocks are computed; the symmetric half is mirrored at
// the write.  The S-dependent O(S*mm^2) work runs on matrix cores; the
// S-independent (mm x mm) solve stays in library fp64.
// ------------------------------------------------------------------------- //

#define GRAM_CHUNK 256
#define GRAM_TJB 4
#define GRAM_MAX_W 8   // M <= 513

template <int W>
__global__ __launch_bounds__(256)
void wls_gram_kernel(
    const uint64_t* __restrict__ packed,  // (B, S, W)
    const float* __restrict__ kw,         // (B, S)
    const float* __restrict__ ey_adj,     // (B, S, n_out)
    const float* __restrict__ total,      // (B, n_out)
    double* __restrict__ A64,             // (B, mm, mm)
    double* __restrict__ rhs64,           // (B, mm, n_out)
    int B, int S, int M, int n_out, int NU)
{
    // one workgroup per (instance, tj-tile-block) work UNIT: a strip-per-ti
    // grid left the chip load-imbalanced (ti=0 owns 4x the tiles of the
    // last strip) and under-filled at 1024 workgroups
    const int b = blockIdx.x / NU;
    int u = blockIdx.x % NU;
    const int mm = M - 1;
    const int cols = mm + n_out;
    const int TJ = (cols + 15) / 16;
    const int TI = (mm + 15) / 16;
    int ti = 0;
    for (; ti < TI; ++ti) {
        const int blocks = (TJ - ti + GRAM_TJB - 1) / GRAM_TJB;
        if (u < blocks) break;
        u -= blocks;
    }
    const int tj0 = ti + u * GRAM_TJB;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wv = tid >> 6;
    const int arow = lane & 15;
    const int akk = lane >> 4;

    extern __shared__ char smem[];
    uint64_t* pk = (uint64_t*)smem;                       // GRAM_CHUNK * W
    float* wch = (float*)(pk + GRAM_CHUNK * W);           // GRAM_CHUNK
    float* mlch = wch + GRAM_CHUNK;                       // GRAM_CHUNK
    float* wmlch = mlch + GRAM_CHUNK;                     // GRAM_CHUNK
    float* eych = wmlch + GRAM_CHUNK;                     // GRAM_CHUNK * n_out
    double* red = (double*)(eych + GRAM_CHUNK * n_out);   // 4 * 16 * 17
    __shared__ float tot_s[WLS_MAX_NOUT];

    if (tid < n_out) tot_s[tid] = total[(size_t)b * n_out + tid];
    __syncthreads();

    const uint64_t* pbase = packed + (size_t)b * S * W;
    const float* kwb = kw + (size_t)b * S;
    const float* eyb = ey_adj + (size_t)b * S * n_out;

    const int i = ti * 16 + arow;          // this lane's Gram row (A operand)
    const int wi = i >> 6, ibit = i & 63;
    const int lw = (M - 1) >> 6, lb = (M - 1) & 63;  // last-mask bit coords
    // per-lane B-column descriptors — loop-invariant across the whole S
    // scan (the in-loop recompute cost 36 VALU per MFMA, PMC-measured)
    int jw[GRAM_TJB], jb[GRAM_TJB], jo[GRAM_TJB];
    bool jgram[GRAM_TJB], jrhs[GRAM_TJB];
#pragma unroll
    for (int tt = 0; tt < GRAM_TJB; ++tt) {
        const int j = (tj0 + tt) * 16 + arow;
        jgram[tt] = j < mm;
        jrhs[tt] = (j >= mm) && (j < cols);
        jw[tt] = (j < mm) ? (j >> 6) : 0;
        jb[tt] = j & 63;
        jo[tt] = j - mm;
    }

    {
        const int ntj = min(GRAM_TJB, TJ - tj0);
        double dacc[GRAM_TJB][4];
        f32x4 acc[GRAM_TJB];
#pragma unroll
        for (int tt = 0; tt < GRAM_TJB; ++tt) {
            acc[tt] = (f32x4){0, 0, 0, 0};
#pragma unroll
            for (int r = 0; r < 4; ++r) dacc[tt][r] = 0.0;
        }

        for (int c0 = 0; c0 < S; c0 += GRAM_CHUNK) {
            const int clen = min(GRAM_CHUNK, S - c0);
            __syncthreads();
            for (int idx = tid; idx < clen * W; idx += 256)
                pk[idx] = pbase[(size_t)(c0 + idx / W) * W + (idx % W)];
            for (int idx = tid; idx < clen; idx += 256) {
                const float w = kwb[c0 + idx];
                const uint64_t lastw = pbase[(size_t)(c0 + idx) * W + lw];
                const float ml = (float)((lastw >> lb) & 1ull);
                wch[idx] = w;
                mlch[idx] = ml;
                wmlch[idx] = w * ml;
                // rhs columns staged pre-weighted: bv is then a pure load
                for (int o = 0; o < n_out; ++o)
                    eych[idx * n_out + o] =
                        w * (eyb[(size_t)(c0 + idx) * n_out + o]
                             - ml * tot_s[o]);
            }
            __syncthreads();
            // wave wv covers samples [wv*64, wv*64+64) of the chunk; the
            // four partial tiles are summed in the cross-wave reduce below
            const int wlo = wv * 64;
            const int whi = min(clen, wlo + 64);
            for (int t4 = wlo; t4 < whi; t4 += 4) {
                const int t = t4 + akk;
                const bool tv = t < whi;
                float a = 0.0f, w = 0.0f, wml = 0.0f;
                const uint64_t* bits = pk + (size_t)t * W;
                if (tv) {
                    w = wch[t];
                    wml = wmlch[t];
                    if (i < mm)
                        a = (((bits[wi] >> ibit) & 1ull) ? 1.0f : 0.0f)
                            - mlch[t];
                }
                // no ntj guard: tiles past TJ have jgram==jrhs==false, so
                // they accumulate exact zeros (the runtime predicate kept
                // the compiler from folding the unrolled body)
#pragma unroll
                for (int tt = 0; tt < GRAM_TJB; ++tt) {
                    float bv = 0.0f;
                    if (tv) {
                        if (jgram[tt])
                            bv = (((bits[jw[tt]] >> jb[tt]) & 1ull)
                                      ? w : 0.0f) - wml;
                        else if (jrhs[tt])
                            bv = eych[t * n_out + jo[tt]];
                    }
                    acc[tt] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                        a, bv, acc[tt], 0, 0, 0);
                }
            }
            // promote the chunk partial to fp64 (fp32 run length <= 256)
#pragma unroll
            for (int tt = 0; tt < GRAM_TJB; ++tt)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    dacc[tt][r] += (double)acc[tt][r];
                    acc[tt][r] = 0.0f;
                }
        }

        // cross-wave reduce + global write, one 16x16 tile at a time
        for (int tt = 0; tt < ntj; ++tt) {
            __syncthreads();
#pragma unroll
            for (int r = 0; r < 4; ++r)
                red[((size_t)wv * 16 + (lane >> 4) * 4 + r) * 17 + (lane & 15)] =
                    dacc[tt][r];
            __syncthreads();
            {
                const int row = tid / 16, col = tid % 16;  // 256 = 16*16
                double v = red[(0 * 16 + row) * 17 + col]
                         + red[(1 * 16 + row) * 17 + col]
                         + red[(2 * 16 + row) * 17 + col]
                         + red[(3 * 16 + row) * 17 + col];
                const int gi = ti * 16 + row;
                const int gj = (tj0 + tt) * 16 + col;
                if (gi < mm) {
                    if (gj < mm) {
                        A64[((size_t)b * mm + gi) * mm + gj] = v;
                        if (gj != gi && gj > gi)
                            A64[((size_t)b * mm + gj) * mm + gi] = v;
                    } else if (gj < cols) {
                        rhs64[((size_t)b * mm + gi) * n_out + (gj - mm)] = v;
                    }
                }
            }
        }
    }
}

extern "C" int launch_wls_gram(
    const uint64_t* packed, const float* kw, const float* ey_adj,
    const float* total, double* A64, double* rhs64, int B, int S, int M,
    int W, int n_out, hipStream_t stream)
{
    if (M < 2 || M > 64 * GRAM_MAX_W + 1 || n_out > WLS_MAX_NOUT || W > GRAM_MAX_W)
        return -1;
    const int mm = M - 1;
    const int TI = (mm + 15) / 16;
    const int TJ = (mm + n_out + 15) / 16;
    int NU = 0;   // upper-triangle tile blocks (rhs cols ride the last ones)
    for (int ti = 0; ti < TI; ++ti)
        NU += (TJ - ti + GRAM_TJB - 1) / GRAM_TJB;
    size_t lds = (size_t)GRAM_CHUNK * W * 8 + (size_t)GRAM_CHUNK * 3 * 4
               + (size_t)GRAM_CHUNK * n_out * 4 + (size_t)4 * 16 * 17 * 8;
    dim3 grid(B * NU), block(256);
#define KSHAP_GRAM_CASE(WV) \
    case WV: \
        wls_gram_kernel<WV><<<grid, block, lds, stream>>>( \
            packed, kw, ey_adj, total, A64, rhs64, B, S, M, n_out, NU); \
        break;
    switch (W) {
        KSHAP_GRAM_CASE(1) KSHAP_GRAM_CASE(2) KSHAP_GRAM_CASE(3)
        KSHAP_GRAM_CASE(4) KSHAP_GRAM_CASE(5) KSHAP_GRAM_CASE(6)
        KSHAP_GRAM_CASE(7) KSHAP_GRAM_CASE(8)
        default: return -1;
    }
#undef KSHAP_GRAM_CASE
    return 0;
}

// ------------------------------------------------------------------------- //
// K3-K6 fused, tiled for the stress shapes (Mpad > 64 or Npad > 128): the
// same mask @ diff MFMA GEMM + activation + weighted background reduction,
// but the diff image is streamed through LDS in (32 k x 128 n) chunks
// instead of staged whole, and each workgroup covers a 128-column background
// tile — its epilogue writes a PARTIAL weighted reduction which a second
// (deterministic, ordered) kernel sums over column tiles into ey.  Grid is
// (b, n-tile, s-tile) with s-tile fastest so consecutive workgroups share a
// diff tile through their XCD's L2.
// ------------------------------------------------------------------------- //

#define FT_NTILE 8          // 16-col sub-tiles per workgroup = 128 columns
#define FT_NSTRIDE 144      // 128 + 16: ≡16 mod 32 -> conflict-free k-pairs

template <int NOUT, int ACT>
__global__ __launch_bounds__(256)
void fused_predict_tiled_kernel(
    const uint8_t* __restrict__ masksU, // (B, S, M)
    const float* __restrict__ diff,     // (B, OIMG, Mpad, Npad)
    const float* __restrict__ base,     // (OIMG, Npad)
    const float* __restrict__ wbg,      // (Npad)  0 for padding cols
    float* __restrict__ partial_out,    // (B, n_ntiles, S, NACC)
    int B, int S, int M, int Mpad, int Npad)
{
    // ACT 3 (binary softmax from the logit difference) accumulates BOTH
    // class sums here — p0 = p1*exp(-z) keeps RELATIVE accuracy for
    // near-saturated probabilities, so the engine's pairwise
    // log(p1)-log(p0) link stays accurate where a (1 - p1) complement
    // would lose everything to fp32 cancellation (stress configs saturate
    // their logits; the reference's fp64 numpy has no such cliff)
    constexpr int OIMG = (ACT == 3) ? 1 : NOUT;
    constexpr int NACC = NOUT;
    // 16-deep k chunks keep the double-buffered LDS footprint ~19 KB
    constexpr int KC = (OIMG >= 4) ? 8 : 16;    // LDS diff chunk k-depth
    constexpr int BUFSZ = OIMG * KC * FT_NSTRIDE;
    constexpr int SREG = OIMG * KC / 2;         // staged elems per thread
    const int n_ntiles = (Npad + 127) / 128;
    const int n_stiles = (S + S_TILE - 1) / S_TILE;
    const int stile = blockIdx.x % n_stiles;
    const int nt = (blockIdx.x / n_stiles) % n_ntiles;
    const int b = blockIdx.x / (n_stiles * n_ntiles);
    const int n0 = nt * 128;
    const int ncols = min(128, Npad - n0);
    const int s0 = stile * S_TILE;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wave = tid >> 6;
    const int swave = wave * 16;
    const int arow = lane & 15;
    const int akcol = lane >> 4;

    extern __shared__ float lds[];
    float* diff_lds = lds;                          // 2 * BUFSZ (ping-pong)
    float* base_lds = diff_lds + 2 * BUFSZ;         // OIMG * 128
    float* wbg_lds = base_lds + OIMG * 128;         // 128

    for (int idx = tid; idx < OIMG * 128; idx += 256) {
        const int o = idx >> 7, n = idx & 127;
        base_lds[idx] = (n < ncols) ? base[(size_t)o * Npad + n0 + n] : 0.0f;
    }
    for (int idx = tid; idx < 128; idx += 256)
        wbg_lds[idx] = (idx < ncols) ? wbg[n0 + idx] : 0.0f;

    const float* dsrc = diff + (size_t)b * OIMG * Mpad * Npad;
    const uint8_t* mlane = masksU + ((size_t)b * S + swave + arow) * M;
    const float* dlane = diff_lds + akcol * FT_NSTRIDE + arow;

    for (int sub = 0; sub < S_TILE / S_SUB; ++sub) {
        const int ssub0 = s0 + sub * S_SUB;
        if (ssub0 >= S) break;
        const int srow = ssub0 + swave + arow;
        const bool svalid = srow < S;
        const uint8_t* mrow = mlane + (size_t)ssub0 * M;

        f32x4 acc[FT_NTILE][OIMG];
#pragma unroll
        for (int ct = 0; ct < FT_NTILE; ++ct)
#pragma unroll
            for (int o = 0; o < OIMG; ++o) acc[ct][o] = (f32x4){0, 0, 0, 0};

        // software-pipelined ping-pong staging: chunk i+1's global loads are
        // issued (into registers) while chunk i's MFMAs run from the other
        // LDS buffer, so the VMEM latency hides and there is ONE barrier per
        // chunk. Staging is strength-reduced: each thread owns one column
        // and one k-parity, addresses stride by 2*Npad (the generic
        // idx-decomposition form cost ~15 VALU/element, PMC).
        const int sn = tid & 127;
        const int sk0 = tid >> 7;            // 0 or 1
        const bool nv = sn < ncols;
        const int nchunks = Mpad / KC;
        float sreg[SREG];
        // preload + write chunk 0 into buffer 0
#pragma unroll
        for (int o = 0; o < OIMG; ++o) {
            const float* srcp = dsrc + ((size_t)o * Mpad + sk0) * Npad + n0 + sn;
#pragma unroll
            for (int q = 0; q < KC / 2; ++q)
                sreg[o * (KC / 2) + q] =
                    nv ? srcp[(size_t)(2 * q) * Npad] : 0.0f;
        }
        __syncthreads();                    // previous sub's reads done
#pragma unroll
        for (int o = 0; o < OIMG; ++o) {
            float* dstp = diff_lds + (o * KC + sk0) * FT_NSTRIDE + sn;
#pragma unroll
            for (int q = 0; q < KC / 2; ++q)
                dstp[2 * q * FT_NSTRIDE] = sreg[o * (KC / 2) + q];
        }

        for (int kci = 0; kci < nchunks; ++kci) {
            const int kc0 = kci * KC;
            const int cur = kci & 1;
            __syncthreads();                 // buffer `cur` writes visible
            if (kci + 1 < nchunks) {
                // issue next chunk's global loads; latency overlaps the MFMAs
#pragma unroll
                for (int o = 0; o < OIMG; ++o) {
                    const float* srcp = dsrc
                        + ((size_t)o * Mpad + kc0 + KC + sk0) * Npad + n0 + sn;
#pragma unroll
                    for (int q = 0; q < KC / 2; ++q)
                        sreg[o * (KC / 2) + q] =
                            nv ? srcp[(size_t)(2 * q) * Npad] : 0.0f;
                }
            }
            // this lane's A bits for the chunk, one batch before the MFMAs
            uint32_t abits = 0;
            if (svalid) {
#pragma unroll
                for (int q = 0; q < KC / 4; ++q) {
                    const int k = kc0 + 4 * q + akcol;
                    abits |= (k < M ? (uint32_t)(mrow[k] & 1) : 0u) << q;
                }
            }
            const float* dl = dlane + cur * BUFSZ;
            // Mpad is a multiple of KC (launcher contract): the k loop fully
            // unrolls and every LDS offset is a compile-time constant
#pragma unroll
            for (int ks = 0; ks < KC; ks += 4) {
                const float a = (float)((abits >> (ks >> 2)) & 1u);
#pragma unroll
                for (int ct = 0; ct < FT_NTILE; ++ct)
#pragma unroll
                    for (int o = 0; o < OIMG; ++o) {
                        const float bv =
                            dl[(o * KC + ks) * FT_NSTRIDE + ct * 16];
                        acc[ct][o] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                            a, bv, acc[ct][o], 0, 0, 0);
                    }
            }
            if (kci + 1 < nchunks) {
                // write next chunk into the other buffer; no barrier needed
                // before the write (everyone passed this iteration's barrier,
                // so no one still reads that buffer)
                float* dst0 = diff_lds + (1 - cur) * BUFSZ;
#pragma unroll
                for (int o = 0; o < OIMG; ++o) {
                    float* dstp = dst0 + (o * KC + sk0) * FT_NSTRIDE + sn;
#pragma unroll
                    for (int q = 0; q < KC / 2; ++q)
                        dstp[2 * q * FT_NSTRIDE] = sreg[o * (KC / 2) + q];
                }
            }
        }

        // epilogue: activation + weighted partial reduction over THIS
        // column tile (cols beyond ncols carry wbg 0 and contribute nothing)
        float partialv[NACC][4];
#pragma unroll
        for (int o = 0; o < NACC; ++o)
#pragma unroll
            for (int r = 0; r < 4; ++r) partialv[o][r] = 0.0f;
#pragma unroll
        for (int ct = 0; ct < FT_NTILE; ++ct) {
            const int n = ct * 16 + arow;
            const float wn = wbg_lds[n];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float z[OIMG];
#pragma unroll
                for (int o = 0; o < OIMG; ++o)
                    z[o] = acc[ct][o][r] + base_lds[o * 128 + n];
                float zz[NACC];
                if (ACT == 3) {
                    const float e = __expf(-z[0]);
                    const float p1 = fast_rcp(1.0f + e);
                    zz[1] = p1;
                    zz[0] = p1 * e;        // sigma(-z): exact-complement sum
                } else if (ACT == 1) {
#pragma unroll
                    for (int o = 0; o < NOUT; ++o)
                        zz[o] = fast_rcp(1.0f + __expf(-z[o]));
                } else if (ACT == 2 && NOUT == 2) {
                    const float e = __expf(z[0] - z[1]);
                    const float p1 = fast_rcp(1.0f + e);
                    zz[0] = p1 * e;
                    zz[1] = p1;
                } else if (ACT == 2) {
                    float mx = z[0];
#pragma unroll
                    for (int o = 1; o < NOUT; ++o) mx = fmaxf(mx, z[o]);
                    float sum = 0.0f;
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) {
                        zz[o] = __expf(z[o] - mx);
                        sum += zz[o];
                    }
                    const float inv = fast_rcp(sum);
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) zz[o] *= inv;
                } else {
#pragma unroll
                    for (int o = 0; o < NACC; ++o) zz[o] = z[o];
                }
#pragma unroll
                for (int o = 0; o < NACC; ++o) partialv[o][r] += wn * zz[o];
            }
        }
#pragma unroll
        for (int o = 0; o < NACC; ++o)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float v = partialv[o][r];
                v += __shfl_xor(v, 1);
                v += __shfl_xor(v, 2);
                v += __shfl_xor(v, 4);
                v += __shfl_xor(v, 8);
                partialv[o][r] = v;
            }
        if (arow == 0) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int ss = ssub0 + swave + akcol * 4 + r;
                if (ss < S)
#pragma unroll
                    for (int o = 0; o < NACC; ++o)
                        partial_out[(((size_t)b * n_ntiles + nt) * S + ss)
                                        * NACC + o] = partialv[o][r];
            }
        }
    }
}

// ------------------------------------------------------------------------- //
// K3-K6 fused, tiled, bf16 matrix cores (opt-in predict_dtype bf16/bf16x2
// for the stress shapes): one v_mfma_f32_16x16x32_bf16 per 32-deep k block
// per column tile — 8x the f32 MFMA rate — with the hi(+lo split) B operand
// staged per k-block from a k-contiguous global image.  Same grid/partial/
// reduce structure as the f32 tiled kernel.
// ------------------------------------------------------------------------- //

#define FTB_KC 32           // k depth of one bf16 MFMA

__global__ void reduce_partials_kernel(
    const float* __restrict__ partial, float* __restrict__ ey,
    size_t total_rows, int S, int n_ntiles, int nacc, int n_out);

template <int NOUT, int ACT, int SPLIT>
__global__ __launch_bounds__(256)
void fused_predict_tiled_bf16_kernel(
    const uint8_t* __restrict__ masksU, // (B, S, M)
    const __bf16* __restrict__ diffB,   // (B, SPLIT, OIMG, Npad, Mpad) k-contig
    const float* __restrict__ base,     // (OIMG, Npad)
    const float* __restrict__ wbg,      // (Npad)
    float* __restrict__ partial_out,    // (B, n_ntiles, S, NACC)
    int B, int S, int M, int Mpad, int Npad)
{
    constexpr int OIMG = (ACT == 3) ? 1 : NOUT;
    constexpr int NACC = NOUT;
    constexpr int BUFSZB = SPLIT * OIMG * 128 * KSTRIDE_BF;  // bf16 elements
    // double-buffered (software-pipelined) staging only for the single-
    // image case (~10 KB/buffer): at SPLIT*OIMG==2 the 41 KB pair capped
    // occupancy at 1 workgroup/CU and measured slower
    constexpr bool DBUF = (SPLIT * OIMG) == 1;
    constexpr int NBUF = DBUF ? 2 : 1;
    const int n_ntiles = (Npad + 127) / 128;
    const int n_stiles = (S + S_TILE - 1) / S_TILE;
    const int stile = blockIdx.x % n_stiles;
    const int nt = (blockIdx.x / n_stiles) % n_ntiles;
    const int b = blockIdx.x / (n_stiles * n_ntiles);
    const int n0 = nt * 128;
    const int ncols = min(128, Npad - n0);
    const int s0 = stile * S_TILE;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wave = tid >> 6;
    const int swave = wave * 16;
    const int arow = lane & 15;          // A row (s) / B col (n)
    const int akb = lane >> 4;           // k-block 0..3 (8 elements each)

    extern __shared__ float lds[];
    __bf16* diff_lds = (__bf16*)lds;     // NBUF * BUFSZB
    float* base_lds = lds + (NBUF * BUFSZB + 1) / 2;
    float* wbg_lds = base_lds + OIMG * 128;

    for (int idx = tid; idx < OIMG * 128; idx += 256) {
        const int o = idx >> 7, n = idx & 127;
        base_lds[idx] = (n < ncols) ? base[(size_t)o * Npad + n0 + n] : 0.0f;
    }
    for (int idx = tid; idx < 128; idx += 256)
        wbg_lds[idx] = (idx < ncols) ? wbg[n0 + idx] : 0.0f;

    const __bf16* dsrc = diffB + (size_t)b * SPLIT * OIMG * Npad * Mpad;
    const uint8_t* mlane = masksU + ((size_t)b * S + swave + arow) * M;
    const __bf16* dlane = diff_lds + (size_t)arow * KSTRIDE_BF + akb * 8;

    for (int sub = 0; sub < S_TILE / S_SUB; ++sub) {
        const int ssub0 = s0 + sub * S_SUB;
        if (ssub0 >= S) break;
        const int srow = ssub0 + swave + arow;
        const bool svalid = srow < S;
        const uint8_t* mrow = mlane + (size_t)ssub0 * M;

        f32x4 acc[FT_NTILE][OIMG];
#pragma unroll
        for (int ct = 0; ct < FT_NTILE; ++ct)
#pragma unroll
            for (int o = 0; o < OIMG; ++o) acc[ct][o] = (f32x4){0, 0, 0, 0};

        // stage the k-block: one thread per (split*o, n) strip copies 32
        // contiguous bf16 from the k-contiguous image into the padded
        // KSTRIDE_BF layout.  With DBUF, chunk i+1's global loads are issued
        // into registers during chunk i's MFMAs (one barrier per chunk).
        const int sso = tid >> 7;
        const int ssn = tid & 127;
        const bool snv = ssn < ncols;
        const int nchunks = Mpad / FTB_KC;
        bf16x8 sreg[FTB_KC / 8];
        if constexpr (DBUF) {
            if (sso < SPLIT * OIMG) {
                const __bf16* sp =
                    dsrc + ((size_t)sso * Npad + n0 + ssn) * Mpad;
#pragma unroll
                for (int q = 0; q < FTB_KC / 8; ++q) {
                    if (snv)
                        sreg[q] = *(const bf16x8*)(sp + q * 8);
                    else
#pragma unroll
                        for (int j = 0; j < 8; ++j) sreg[q][j] = (__bf16)0.0f;
                }
            }
        }
        __syncthreads();                 // prior sub's reads complete
        if constexpr (DBUF) {
            if (sso < SPLIT * OIMG) {
                __bf16* dp =
                    diff_lds + ((size_t)sso * 128 + ssn) * KSTRIDE_BF;
#pragma unroll
                for (int q = 0; q < FTB_KC / 8; ++q)
                    *(bf16x8*)(dp + q * 8) = sreg[q];
            }
        }

        for (int kci = 0; kci < nchunks; ++kci) {
            const int kc0 = kci * FTB_KC;
            const int cur = DBUF ? (kci & 1) : 0;
            if constexpr (DBUF) {
                __syncthreads();         // buffer `cur` writes visible
                if (kci + 1 < nchunks && sso < SPLIT * OIMG) {
                    const __bf16* sp = dsrc
                        + ((size_t)sso * Npad + n0 + ssn) * Mpad + kc0 + FTB_KC;
#pragma unroll
                    for (int q = 0; q < FTB_KC / 8; ++q) {
                        if (snv)
                            sreg[q] = *(const bf16x8*)(sp + q * 8);
                        else
#pragma unroll
                            for (int j = 0; j < 8; ++j)
                                sreg[q][j] = (__bf16)0.0f;
                    }
                }
            } else {
                __syncthreads();
                for (int so = sso; so < SPLIT * OIMG; so += 2) {
                    const __bf16* sp =
                        dsrc + ((size_t)so * Npad + n0 + ssn) * Mpad + kc0;
                    __bf16* dp =
                        diff_lds + ((size_t)so * 128 + ssn) * KSTRIDE_BF;
#pragma unroll
                    for (int q = 0; q < FTB_KC / 8; ++q) {
                        bf16x8 v;
                        if (snv)
                            v = *(const bf16x8*)(sp + q * 8);
                        else
#pragma unroll
                            for (int j = 0; j < 8; ++j)
                                v[j] = (__bf16)0.0f;
                        *(bf16x8*)(dp + q * 8) = v;
                    }
                }
                __syncthreads();
            }
            // A fragment: this lane's 8 mask bits for the k-block, converted
            // in-register (exact in bf16)
            bf16x8 a;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int k = kc0 + akb * 8 + j;
                a[j] = (__bf16)(float)(
                    (svalid && k < M) ? (mrow[k] & 1) : 0);
            }
            const __bf16* dl = dlane + (size_t)cur * BUFSZB;
#pragma unroll
            for (int ct = 0; ct < FT_NTILE; ++ct)
#pragma unroll
                for (int o = 0; o < OIMG; ++o)
#pragma unroll
                    for (int sp = 0; sp < SPLIT; ++sp) {
                        const bf16x8 bv = *(const bf16x8*)(
                            dl + ((size_t)(sp * OIMG + o) * 128 + ct * 16)
                                     * KSTRIDE_BF);
                        acc[ct][o] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a, bv, acc[ct][o], 0, 0, 0);
                    }
            if constexpr (DBUF) {
                if (kci + 1 < nchunks && sso < SPLIT * OIMG) {
                    __bf16* dp = diff_lds + (size_t)(1 - cur) * BUFSZB
                                 + ((size_t)sso * 128 + ssn) * KSTRIDE_BF;
#pragma unroll
                    for (int q = 0; q < FTB_KC / 8; ++q)
                        *(bf16x8*)(dp + q * 8) = sreg[q];
                }
            }
        }

        float partialv[NACC][4];
#pragma unroll
        for (int o = 0; o < NACC; ++o)
#pragma unroll
            for (int r = 0; r < 4; ++r) partialv[o][r] = 0.0f;
#pragma unroll
        for (int ct = 0; ct < FT_NTILE; ++ct) {
            const int n = ct * 16 + arow;
            const float wn = wbg_lds[n];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float z[OIMG];
#pragma unroll
                for (int o = 0; o < OIMG; ++o)
                    z[o] = acc[ct][o][r] + base_lds[o * 128 + n];
                float zz[NACC];
                if (ACT == 3) {
                    const float e = __expf(-z[0]);
                    const float p1 = fast_rcp(1.0f + e);
                    zz[1] = p1;
                    zz[0] = p1 * e;
                } else if (ACT == 1) {
#pragma unroll
                    for (int o = 0; o < NOUT; ++o)
                        zz[o] = fast_rcp(1.0f + __expf(-z[o]));
                } else if (ACT == 2 && NOUT == 2) {
                    const float e = __expf(z[0] - z[1]);
                    const float p1 = fast_rcp(1.0f + e);
                    zz[0] = p1 * e;
                    zz[1] = p1;
                } else if (ACT == 2) {
                    float mx = z[0];
#pragma unroll
                    for (int o = 1; o < NOUT; ++o) mx = fmaxf(mx, z[o]);
                    float sum = 0.0f;
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) {
                        zz[o] = __expf(z[o] - mx);
                        sum += zz[o];
                    }
                    const float inv = fast_rcp(sum);
#pragma unroll
                    for (int o = 0; o < NOUT; ++o) zz[o] *= inv;
                } else {
#pragma unroll
                    for (int o = 0; o < NACC; ++o) zz[o] = z[o];
                }
#pragma unroll
                for (int o = 0; o < NACC; ++o) partialv[o][r] += wn * zz[o];
            }
        }
#pragma unroll
        for (int o = 0; o < NACC; ++o)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float v = partialv[o][r];
                v += __shfl_xor(v, 1);
                v += __shfl_xor(v, 2);
                v += __shfl_xor(v, 4);
                v += __shfl_xor(v, 8);
                partialv[o][r] = v;
            }
        if (arow == 0) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int ss = ssub0 + swave + akb * 4 + r;
                if (ss < S)
#pragma unroll
                    for (int o = 0; o < NACC; ++o)
                        partial_out[(((size_t)b * n_ntiles + nt) * S + ss)
                                        * NACC + o] = partialv[o][r];
            }
        }
    }
}

__global__ void build_diff_bf16_tiled_kernel(
    const float* __restrict__ xp,    // (B, G, O)
    const float* __restrict__ bgp,   // (N, G, O)
    const int64_t* __restrict__ vidx,  // (m,)
    __bf16* __restrict__ out,        // (B, SPLIT, O, Npad, Mpad) k-contig
    int G, int O, int N, int m, int Mpad, int Npad, int split, size_t total)
{
    size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;       // total = B*O*N*m, k fastest
    const int k = idx % m;
    const int n = (idx / m) % N;
    const int o = (idx / ((size_t)m * N)) % O;
    const size_t b = idx / ((size_t)m * N * O);
    const int64_t g = vidx[k];
    const float v = xp[((size_t)b * G + g) * O + o]
                  - bgp[((size_t)n * G + g) * O + o];
    const __bf16 hi = (__bf16)v;
    const size_t basei =
        ((((size_t)b * split) * O + o) * Npad + n) * Mpad + k;
    out[basei] = hi;
    if (split == 2) {
        const size_t lo =
            ((((size_t)b * split + 1) * O + o) * Npad + n) * Mpad + k;
        out[lo] = (__bf16)(v - (float)hi);
    }
}

extern "C" void launch_build_diff_bf16_tiled(
    const float* xp, const float* bgp, const int64_t* vidx, uint16_t* out,
    int B, int G, int O, int N, int m, int Mpad, int Npad, int split,
    hipStream_t stream)
{
    size_t total = (size_t)B * O * N * m;
    build_diff_bf16_tiled_kernel<<<dim3((unsigned)((total + 255) / 256)),
                                   dim3(256), 0, stream>>>(
        xp, bgp, vidx, reinterpret_cast<__bf16*>(out), G, O, N, m, Mpad,
        Npad, split, total);
}

template <int NOUT, int ACT, int SPLIT>
static void launch_ftb_one(
    const uint8_t* masksU, const __bf16* diffB, const float* base,
    const float* wbg, float* partial, float* ey, int B, int S, int M,
    int Mpad, int Npad, hipStream_t stream)
{
    constexpr int OIMG = (ACT == 3) ? 1 : NOUT;
    constexpr int NACC = NOUT;
    constexpr int NBUF = ((SPLIT * OIMG) == 1) ? 2 : 1;  // mirror DBUF
    const int n_ntiles = (Npad + 127) / 128;
    const int n_stiles = (S + S_TILE - 1) / S_TILE;
    dim3 grid(B * n_ntiles * n_stiles), block(256);
    size_t lds = ((size_t)NBUF * SPLIT * OIMG * 128 * KSTRIDE_BF * 2 + 2)
               + (size_t)(OIMG * 128 + 128) * 4 + 4;
    fused_predict_tiled_bf16_kernel<NOUT, ACT, SPLIT>
        <<<grid, block, lds, stream>>>(
            masksU, diffB, base, wbg, partial, B, S, M, Mpad, Npad);
    size_t rows = (size_t)B * S;
    reduce_partials_kernel<<<dim3((unsigned)((rows + 255) / 256)), dim3(256),
                             0, stream>>>(
        partial, ey, rows, S, n_ntiles, NACC, NOUT);
}

template <int NOUT, int ACT>
static void launch_ftb_split(
    const uint8_t* masksU, const __bf16* diffB, const float* base,
    const float* wbg, float* partial, float* ey, int B, int S, int M,
    int Mpad, int Npad, int split, hipStream_t stream)
{
    if (split == 2)
        launch_ftb_one<NOUT, ACT, 2>(masksU, diffB, base, wbg, partial, ey,
                                     B, S, M, Mpad, Npad, stream);
    else
        launch_ftb_one<NOUT, ACT, 1>(masksU, diffB, base, wbg, partial, ey,
                                     B, S, M, Mpad, Npad, stream);
}

template <int NOUT>
static void launch_ftb_act(
    const uint8_t* masksU, const __bf16* diffB, const float* base,
    const float* wbg, float* partial, float* ey, int B, int S, int M,
    int Mpad, int Npad, int act, int split, hipStream_t stream)
{
    switch (act) {
        case 0:
            launch_ftb_split<NOUT, 0>(masksU, diffB, base, wbg, partial, ey, B, S, M, Mpad, Npad, split, stream);
            break;
        case 1:
            launch_ftb_split<NOUT, 1>(masksU, diffB, base, wbg, partial, ey, B, S, M, Mpad, Npad, split, stream);
            break;
        case 3:
            if constexpr (NOUT == 2)
                launch_ftb_split<NOUT, 3>(masksU, diffB, base, wbg, partial, ey, B, S, M, Mpad, Npad, split, stream);
            break;
        default:
            launch_ftb_split<NOUT, 2>(masksU, diffB, base, wbg, partial, ey, B, S, M, Mpad, Npad, split, stream);
            break;
    }
}

extern "C" int launch_fused_predict_tiled_bf16(
    const uint8_t* masksU, const uint16_t* diffB_u, const float* base,
    const float* wbg, float* partial, float* ey, int B, int S, int M,
    int Mpad, int Npad, int n_out, int act, int split, hipStream_t stream)
{
    if (Npad % 16 != 0 || Mpad % FTB_KC != 0 || split < 1 || split > 2)
        return -1;
    const __bf16* diffB = reinterpret_cast<const __bf16*>(diffB_u);
    switch (n_out) {
        case 1: launch_ftb_act<1>(masksU, diffB, base, wbg, partial, ey, B, S, M, Mpad, Npad, act, split, stream); break;
        case 2: launch_ftb_act<2>(masksU, diffB, base, wbg, partial, ey, B, S, M, Mpad, Npad, act, split, stream); break;
        case 4: launch_ftb_act<4>(masksU, diffB, base, wbg, partial, ey, B, S, M, Mpad, Npad, act, split, stream); break;
        default: return -1;
    }
    return 0;
}

// deterministic column-tile reduction: ey[b,s,:] from the per-tile partials
// (a fixed summation order — float atomics would break the bitwise-
// determinism guarantee the GPU tests assert)
__global__ void reduce_partials_kernel(
    const float* __restrict__ partial,  // (B, n_ntiles, S, nacc)
    float* __restrict__ ey,             // (B, S, n_out)
    size_t total_rows, int S, int n_ntiles, int nacc, int n_out)
{
    const size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total_rows) return;       // total_rows = B * S
    const size_t b = idx / S;
    const int s = (int)(idx % S);
    for (int o = 0; o < nacc; ++o) {
        float v = 0.0f;
        for (int nt = 0; nt < n_ntiles; ++nt)
            v += partial[(((size_t)b * n_ntiles + nt) * S + s) * nacc + o];
        ey[((size_t)b * S + s) * n_out + o] = v;
    }
}

template <int NOUT, int ACT>
static void launch_ft(
    const uint8_t* masksU, const float* diff, const float* base,
    const float* wbg, float* partial, float* ey, int B, int S, int M,
    int Mpad, int Npad, hipStream_t stream)
{
    constexpr int OIMG = (ACT == 3) ? 1 : NOUT;
    constexpr int NACC = NOUT;
    constexpr int KC = (OIMG >= 4) ? 8 : 16;
    const int n_ntiles = (Npad + 127) / 128;
    const int n_stiles = (S + S_TILE - 1) / S_TILE;
    dim3 grid(B * n_ntiles * n_stiles), block(256);
    size_t lds = (size_t)(2 * OIMG * KC * FT_NSTRIDE + OIMG * 128 + 128) * 4;
    fused_predict_tiled_kernel<NOUT, ACT><<<grid, block, lds, stream>>>(
        masksU, diff, base, wbg, partial, B, S, M, Mpad, Npad);
    size_t rows = (size_t)B * S;
    reduce_partials_kernel<<<dim3((unsigned)((rows + 255) / 256)), dim3(256),
                             0, stream>>>(
        partial, ey, rows, S, n_ntiles, NACC, NOUT);
}

template <int NOUT>
static void launch_ft_act(
    const uint8_t* masksU, const float* diff, const float* base,
    const float* wbg, float* partial, float* ey, int B, int S, int M,
    int Mpad, int Npad, int act, hipStream_t stream)
{
    switch (act) {
        case 0:
            launch_ft<NOUT, 0>(masksU, diff, base, wbg, partial, ey, B, S, M, Mpad, Npad, stream);
            break;
        case 1:
            launch_ft<NOUT, 1>(masksU, diff, base, wbg, partial, ey, B, S, M, Mpad, Npad, stream);
            break;
        case 3:
            if constexpr (NOUT == 2)
                launch_ft<NOUT, 3>(masksU, diff, base, wbg, partial, ey, B, S, M, Mpad, Npad, stream);
            break;
        default:
            launch_ft<NOUT, 2>(masksU, diff, base, wbg, partial, ey, B, S, M, Mpad, Npad, stream);
            break;
    }
}

extern "C" int launch_fused_predict_tiled(
    const uint8_t* masksU, const float* diff, const float* base,
    const float* wbg, float* partial, float* ey, int B, int S, int M,
    int Mpad, int Npad, int n_out, int act, hipStream_t stream)
{
    if (Npad % 16 != 0 || Mpad % 16 != 0) return -1;  // Mpad % KC == 0
    switch (n_out) {
        case 1: launch_ft_act<1>(masksU, diff, base, wbg, partial, ey, B, S, M, Mpad, Npad, act, stream); break;
        case 2: launch_ft_act<2>(masksU, diff, base, wbg, partial, ey, B, S, M, Mpad, Npad, act, stream); break;
        case 4: launch_ft_act<4>(masksU, diff, base, wbg, partial, ey, B, S, M, Mpad, Npad, act, stream); break;
        default: return -1;
    }
    return 0;
}
