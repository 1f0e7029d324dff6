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

#define MLP_BM 32          // rows per block
#define MLP_HMAX 512       // max hidden width staged in LDS

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
    const int half = wave & 1;
    const int cstart = wave >> 1;               // 0..3
    for (long c = cstart * 16; c < H; c += 64) {
        f32x4 acc = tile_16x16(A, lda, half * 16, W, K, c, K, lane);
        const int col = lane & 15;
        float bv = (float)bias[c + col];
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int row = (lane >> 4) * 4 + r;
            float v = acc[r] + bv;
            v = v > 0.f ? v : 0.f;
            mbf16 hv = (mbf16)v;
            lds_out[(half * 16 + row) * MLP_HMAX + c + col] = hv;
            long gm = m0_global + half * 16 + row;
            if (gm < M) save[gm * H + c + col] = hv;
        }
    }
}

extern "C" __global__ __launch_bounds__(512, 1)
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
    {
        const int half = wave & 1;
        const int cstart = wave >> 1;
        for (long c = cstart * 16; c < H; c += 64) {
            f32x4 acc = tile_16x16(x0, K0, m0 + half * 16, w1, K0, c, K0,
                                   lane);
            const int col = lane & 15;
            float bv = (float)b1[c + col];
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = (lane >> 4) * 4 + r;
                float v = acc[r] + bv;
                v = v > 0.f ? v : 0.f;
                mbf16 hv = (mbf16)v;
                act[0][(half * 16 + row) * MLP_HMAX + c + col] = hv;
                long gm = m0 + half * 16 + row;
                if (gm < M) a1[gm * H + c + col] = hv;
            }
        }
    }
    __syncthreads();
    layer(act[0], MLP_HMAX, m0, w2, b2, H, H, act[1], a2, M, wave, lane);
    __syncthreads();
    layer(act[1], MLP_HMAX, m0, w3, b3, H, H, act[0], a3, M, wave, lane);
    __syncthreads();

    // final Linear(H, 1): out[m] = A3[m] . w4 + b4, VALU reduction.
    // wave w handles rows w*4 .. w*4+3 of the 32-row tile.
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

extern "C" void emb_mlp3_fwd(const void* x0, long M, long K0,
                             const void* w1, const void* b1,
                             const void* w2, const void* b2,
                             const void* w3, const void* b3,
                             const void* w4, const void* b4, long H,
                             void* a1, void* a2, void* a3, float* out,
                             hipStream_t stream) {
    if (M == 0) return;
    long grid = (M + MLP_BM - 1) / MLP_BM;
    k_mlp3_fwd<<<(int)grid, 512, 0, stream>>>(
        (const mbf16*)x0, M, K0, (const mbf16*)w1, (const mbf16*)b1,
        (const mbf16*)w2, (const mbf16*)b2, (const mbf16*)w3,
        (const mbf16*)b3, (const mbf16*)w4, (const mbf16*)b4, H,
        (mbf16*)a1, (mbf16*)a2, (mbf16*)a3, out);
}
