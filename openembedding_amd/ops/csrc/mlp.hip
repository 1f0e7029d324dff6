// Fused 3-hidden-layer MLP forward for the CTR DNN (gfx950, bf16 MFMA).
//
// The DeepFM/WDL dnn is Linear(K0,H)+ReLU ×3 then Linear(H,1). At M=4096,
// H=400 these GEMMs are too skinny for the library path (hipBLASLt fp32
// measured ~43 TF, ~26 us per layer, plus separate bias/ReLU/cast
// launches). This kernel runs the WHOLE forward in one launch:
// layer-to-layer activations stay in LDS, weights stream from L2/L3
// (~1 MB bf16, resident), MFMA 16x16x32 bf16 tiles, bias+ReLU fused in the
// epilogue. Hidden activations are also written to HBM (bf16) because the
// backward needs them (ReLU mask = act > 0).
//
// Shapes: X0 [M, K0] bf16 (K0 arbitrary), W_l stored torch-Linear style
// [out, in] bf16 (so the MFMA B-fragment "B[k][n] = W[n][k]" is 8
// contiguous k per lane = one 16-byte load), H multiple of 16, final
// W4 [1, H] + b4 -> out [M] fp32.
//
// Fragment maps (mfma_f32_16x16x32_bf16, verified by the numerics test
// vs a torch fp32 reference with asymmetric inputs):
//   A[m][k]:  m = lane&15, k = (lane>>4)*8 + e   (e = 0..7)
//   B[k][n]:  n = lane&15, k = (lane>>4)*8 + e
//   C/D:      col = lane&15, row = (lane>>4)*4 + r  (r = 0..3)

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __hip_bfloat16 mbf16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define MLP_BM 16          // rows per block (one 16-row MFMA tile)
#define MLP_HMAX 408       // LDS row stride: 408*2B = 204 dwords, 204 % 64
                           // = 12 -> the 16 rows of a fragment read hit 16
                           // distinct banks (12*m mod 64 has period 16)

// load an 8-element bf16 A/B fragment from row-major [rows, ld] at
// (row, kbase..kbase+7), zero-filling past K (ragged tails: K0=247, H=400)
static __device__ __forceinline__ bf16x8 frag_row(const mbf16* base, long ld,
                                                  long row, long kbase,
                                                  long K) {
    const mbf16* p = base + row * ld + kbase;
    bf16x8 f;
    if (kbase + 8 <= K) {
        f = *reinterpret_cast<const bf16x8*>(p);  // 16-byte load
    } else {
        #pragma unroll
        for (int e = 0; e < 8; ++e)
            f[e] = (kbase + e < K) ? (__bf16)p[e] : (__bf16)0.0f;
    }
    return f;
}

// one 16x16 output tile: rows [m0,m0+16) of act_in vs cols [n0,n0+16) of W
static __device__ __forceinline__ f32x4 tile_16x16(
        const mbf16* A, long lda, long m0,
        const mbf16* W, long ldw, long n0, long K, int lane) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const long m = m0 + (lane & 15);
    const long n = n0 + (lane & 15);
    const long koff = (lane >> 4) * 8;
    for (long kb = 0; kb < K; kb += 32) {
        bf16x8 a = frag_row(A, lda, m, kb + koff, K);
        bf16x8 b = frag_row(W, ldw, n, kb + koff, K);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    return acc;
}

// Layer loop body: computes act_out[0:BM][0:H] = relu(act_in @ W^T + b)
// into LDS (ld = MLP_HMAX) and mirrors it to HBM save buffer [M, H].
// 8 waves: wave&1 selects the 16-row half, wave>>1 strides the col chunks.
static __device__ __forceinline__ void layer(
        const mbf16* A, long lda,      // input rows base (global or LDS)
        long m0_global,                // for the HBM mirror
        const mbf16* W, const mbf16* bias, long H, long K,
        mbf16* lds_out, mbf16* save, long M,
        int wave, int lane) {
    for (long c = wave * 16; c < H; c += 64) {
        f32x4 acc = tile_16x16(A, lda, 0, W, K, c, K, lane);
        const int col = lane & 15;
        float bv = (float)bias[c + col];
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int row = (lane >> 4) * 4 + r;
            float v = acc[r] + bv;
            v = v > 0.f ? v : 0.f;
            mbf16 hv = (mbf16)v;
            lds_out[row * MLP_HMAX + c + col] = hv;
            long gm = m0_global + row;
            if (gm < M) save[gm * H + c + col] = hv;
        }
    }
}

extern "C" __global__ __launch_bounds__(256, 2)
void k_mlp3_fwd(const mbf16* __restrict__ x0, long M, long K0,
                const mbf16* __restrict__ w1, const mbf16* __restrict__ b1,
                const mbf16* __restrict__ w2, const mbf16* __restrict__ b2,
                const mbf16* __restrict__ w3, const mbf16* __restrict__ b3,
                const mbf16* __restrict__ w4, const mbf16* __restrict__ b4,
                long H,
                mbf16* __restrict__ a1, mbf16* __restrict__ a2,
                mbf16* __restrict__ a3, float* __restrict__ out) {
    __shared__ mbf16 act[2][MLP_BM * MLP_HMAX];
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const long m0 = (long)blockIdx.x * MLP_BM;
    if (m0 >= M) return;

    // layer 1: read X0 straight from global (row-major, 16B fragments)
    for (long c = wave * 16; c < H; c += 64) {
        f32x4 acc = tile_16x16(x0, K0, m0, w1, K0, c, K0, lane);
        const int col = lane & 15;
        float bv = (float)b1[c + col];
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int row = (lane >> 4) * 4 + r;
            float v = acc[r] + bv;
            v = v > 0.f ? v : 0.f;
            mbf16 hv = (mbf16)v;
            act[0][row * MLP_HMAX + c + col] = hv;
            long gm = m0 + row;
            if (gm < M) a1[gm * H + c + col] = hv;
        }
    }
    __syncthreads();
    layer(act[0], MLP_HMAX, m0, w2, b2, H, H, act[1], a2, M, wave, lane);
    __syncthreads();
    layer(act[1], MLP_HMAX, m0, w3, b3, H, H, act[0], a3, M, wave, lane);
    __syncthreads();

    // final Linear(H, 1): out[m] = A3[m] . w4 + b4, VALU reduction.
    // wave w handles rows w*4 .. w*4+3 of the 16-row tile.
    for (int r = 0; r < 4; ++r) {
        long row = wave * 4 + r;
        long gm = m0 + row;
        if (gm >= M) continue;
        float s = 0.f;
        for (long k = lane; k < H; k += 64)
            s += (float)act[0][row * MLP_HMAX + k] * (float)w4[k];
        #pragma unroll
        for (int off = 32; off; off >>= 1)
            s += __shfl_down(s, off, 64);
        if (lane == 0) out[gm] = s + (float)b4[0];
    }
}

// ---------------------------------------------------------------- backward
// Fused dgrad chain: dz3 = dout*w4 ⊙ relu'(a3); dz2 = dz3@W3 ⊙ relu'(a2);
// dz1 = dz2@W2 ⊙ relu'(a1); dx0 = dz1@W1. The GEMMs take the TRANSPOSED
// weights (WnT rows contiguous in the contraction dim, prepared host-side)
// so the B fragment stays one 16-byte load. dz tiles flow through LDS;
// dz1..3 are also written to HBM for the (library-friendly, K=M) wgrad
// GEMMs. relu' masks read the saved activations.

// GEMM layer of the chain: lds_out/save <- (A_lds @ WT) ⊙ mask(act>0)
static __device__ __forceinline__ void bwd_layer(
        const mbf16* A_lds,            // [16, MLP_HMAX] LDS
        long m0_global,
        const mbf16* WT, long N, long K,   // WT [N, K] rows contiguous
        const mbf16* act, long act_ld,     // relu mask source or nullptr
        mbf16* lds_out, mbf16* save, long save_ld, long M,
        int wave, int lane) {
    for (long c = wave * 16; c < N; c += 64) {
        f32x4 acc = tile_16x16(A_lds, MLP_HMAX, 0, WT, K, c, K, lane);
        const int col = lane & 15;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int row = (lane >> 4) * 4 + r;
            long gm = m0_global + row;
            float v = acc[r];
            if (act && gm < M
                && !((float)act[gm * act_ld + c + col] > 0.f))
                v = 0.f;
            mbf16 hv = (mbf16)v;
            if (lds_out) lds_out[row * MLP_HMAX + c + col] = hv;
            if (gm < M) save[gm * save_ld + c + col] = hv;
        }
    }
}

extern "C" __global__ __launch_bounds__(256, 2)
void k_mlp3_bwd(const float* __restrict__ dout, long M, long K0p,
                const mbf16* __restrict__ a1, const mbf16* __restrict__ a2,
                const mbf16* __restrict__ a3,
                const mbf16* __restrict__ w4,    // [H]
                const mbf16* __restrict__ w3t,   // [H, H]
                const mbf16* __restrict__ w2t,   // [H, H]
                const mbf16* __restrict__ w1t,   // [K0p, H]
                long H,
                mbf16* __restrict__ dz1, mbf16* __restrict__ dz2,
                mbf16* __restrict__ dz3, mbf16* __restrict__ dx0) {
    __shared__ mbf16 dz[2][MLP_BM * MLP_HMAX];
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const long m0 = (long)blockIdx.x * MLP_BM;
    if (m0 >= M) return;

    // dz3: elementwise outer product with relu mask
    for (long i = threadIdx.x; i < MLP_BM * H; i += blockDim.x) {
        long row = i / H, n = i % H;
        long gm = m0 + row;
        float v = 0.f;
        if (gm < M && (float)a3[gm * H + n] > 0.f)
            v = dout[gm] * (float)w4[n];
        mbf16 hv = (mbf16)v;
        dz[0][row * MLP_HMAX + n] = hv;
        if (gm < M) dz3[gm * H + n] = hv;
    }
    __syncthreads();
    bwd_layer(dz[0], m0, w3t, H, H, a2, H, dz[1], dz2, H, M, wave, lane);
    __syncthreads();
    bwd_layer(dz[1], m0, w2t, H, H, a1, H, dz[0], dz1, H, M, wave, lane);
    __syncthreads();
    bwd_layer(dz[0], m0, w1t, K0p, H, nullptr, 0, nullptr, dx0, K0p, M,
              wave, lane);
}

extern "C" void emb_mlp3_bwd(const float* dout, long M, long K0p,
                             const void* a1, const void* a2, const void* a3,
                             const void* w4, const void* w3t,
                             const void* w2t, const void* w1t, long H,
                             void* dz1, void* dz2, void* dz3, void* dx0,
                             hipStream_t stream) {
    if (M == 0) return;
    long grid = (M + MLP_BM - 1) / MLP_BM;
    k_mlp3_bwd<<<(int)grid, 256, 0, stream>>>(
        dout, M, K0p, (const mbf16*)a1, (const mbf16*)a2, (const mbf16*)a3,
        (const mbf16*)w4, (const mbf16*)w3t, (const mbf16*)w2t,
        (const mbf16*)w1t, H, (mbf16*)dz1, (mbf16*)dz2, (mbf16*)dz3,
        (mbf16*)dx0);
}

extern "C" void emb_mlp3_fwd(const void* x0, long M, long K0,
                             const void* w1, const void* b1,
                             const void* w2, const void* b2,
                             const void* w3, const void* b3,
                             const void* w4, const void* b4, long H,
                             void* a1, void* a2, void* a3, float* out,
                             hipStream_t stream) {
    if (M == 0) return;
    long grid = (M + MLP_BM - 1) / MLP_BM;
    k_mlp3_fwd<<<(int)grid, 256, 0, stream>>>(
        (const mbf16*)x0, M, K0, (const mbf16*)w1, (const mbf16*)b1,
        (const mbf16*)w2, (const mbf16*)b2, (const mbf16*)w3,
        (const mbf16*)b3, (const mbf16*)w4, (const mbf16*)b4, H,
        (mbf16*)a1, (mbf16*)a2, (mbf16*)a3, out);
}
