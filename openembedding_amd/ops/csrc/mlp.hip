// Fused 3-hidden-layer MLP (gfx950, bf16 MFMA 16x16x32) — forward and the
// dgrad-chain backward, each in ONE kernel.
//
// Why: at M=4096, H=400 the per-layer library GEMMs are skinny (measured
// ~43 TF fp32, ~26 us each) and the bias/ReLU/cast glue adds ~20 more
// launches. Here activations flow layer-to-layer through LDS; weights
// stream from L2 (1 MB bf16, L2-resident per XCD); bias+ReLU fuse into the
// MFMA epilogue; hidden activations mirror to HBM for the backward.
//
// EVERY contraction length is padded to a multiple of 32 host-side (padded
// weight copies with zeroed tails; x0 zero-padded by the head kernel): the
// fragment load is then one unconditional 16-byte load, which is what lets
// hipcc software-pipeline the k-loop (the earlier bounds-checked version
// scalarized to 2-byte loads and serialized load->mfma per k-step:
// 93 us/step; see git history).
//
// Fragment maps (mfma_f32_16x16x32_bf16, verified by tests/test_gpu_mlp.py
// against a torch reference with asymmetric data):
//   A[m][k]: m = lane&15, k = (lane>>4)*8 + e
//   B[k][n]: n = lane&15, k = (lane>>4)*8 + e   (B = W[n][k] row-major)
//   C/D:     col = lane&15, row = (lane>>4)*4 + r

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __hip_bfloat16 mbf16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define MLP_BM 16    // rows per block = one MFMA row tile
// LDS row stride (elements): 424*2B = 212 dwords; 212 % 64 = 20,
// gcd(20,64)=4 -> the 16 rows of a fragment read land on 16 distinct banks.
#define MLP_LD 424

static __device__ __forceinline__ bf16x8 ld_frag(const mbf16* p) {
    return *reinterpret_cast<const bf16x8*>(
        __builtin_assume_aligned(p, 16));
}

// 16x16 tile over a padded contraction: A [.., lda] row-major, W [N, ldw]
// row-major (B = W[n][k]); Kp % 32 == 0.
//
// M=16 per block means the W stream has NO reuse inside the block — the
// "decode-GEMV" regime of the perf guide: load straight to VGPRs with a
// DEEP UNROLL and late counted waits, so all K-steps' loads are in flight
// before the first MFMA needs its operands (a 2-deep pipeline left ~200
// cycles of L2 latency exposed per step: 80-90 us/kernel measured).
template <int KS>
static __device__ __forceinline__ f32x4 tile16_u(
        const mbf16* __restrict__ pa, const mbf16* __restrict__ pb) {
    bf16x8 a[KS], b[KS];
    #pragma unroll
    for (int s = 0; s < KS; ++s) {
        a[s] = ld_frag(pa + 32 * s);
        b[s] = ld_frag(pb + 32 * s);
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int s = 0; s < KS; ++s)
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[s], b[s], acc,
                                                      0, 0, 0);
    return acc;
}

static __device__ __forceinline__ f32x4 tile16(
        const mbf16* A, long lda, const mbf16* W, long ldw, long n0,
        long Kp, int lane) {
    const long koff = (lane >> 4) * 8;
    const mbf16* pa = A + (lane & 15) * lda + koff;
    const mbf16* pb = W + (n0 + (lane & 15)) * ldw + koff;
    if (Kp == 256) return tile16_u<8>(pa, pb);    // K0 = 247 padded
    if (Kp == 416) return tile16_u<13>(pa, pb);   // H = 400 padded
    if (Kp == 128) return tile16_u<4>(pa, pb);
    // generic fallback: 2-stage pipeline
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    bf16x8 a0 = ld_frag(pa), b0 = ld_frag(pb);
    for (long kb = 32; kb < Kp; kb += 32) {
        bf16x8 a1 = ld_frag(pa + kb);
        bf16x8 b1 = ld_frag(pb + kb);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
        a0 = a1; b0 = b1;
    }
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
}

#define MLP_WAVES 8   // 8 waves/block = 2 per SIMD: the partner wave hides
                      // the other's W-stream latency (1 block/CU grid)

// one layer: lds_out/save <- relu(A @ W^T + b) (relu/bias optional).
//
// Pipelining (PMC showed 86% SQ_WAIT_ANY with per-chunk batched loads):
// the A fragments depend only on (row, k) — identical for every column
// chunk — so they load ONCE per layer into registers; the B (weight)
// stream double-buffers across chunks: chunk i+1's 13 fragment loads are
// in flight while chunk i's MFMAs run.
template <int KS>
static __device__ __forceinline__ void mlp_layer_u(
        const mbf16* A, long lda, long m0, long M,
        const mbf16* W, long ldw, const mbf16* bias, long H,
        const mbf16* mask_act, long mask_ld,
        mbf16* lds_out, mbf16* save, long save_ld,
        int wave, int lane, bool relu) {
    const long koff = (lane >> 4) * 8;
    const mbf16* pa = A + (lane & 15) * lda + koff;
    bf16x8 a[KS];
    #pragma unroll
    for (int s = 0; s < KS; ++s) a[s] = ld_frag(pa + 32 * s);

    const int col = lane & 15;
    bf16x8 b0[KS], b1[KS];
    long c = wave * 16;
    if (c < H) {
        const mbf16* pb = W + (c + col) * ldw + koff;
        #pragma unroll
        for (int s = 0; s < KS; ++s) b0[s] = ld_frag(pb + 32 * s);
    }
    for (; c < H; c += 16 * MLP_WAVES) {
        const long cn = c + 16 * MLP_WAVES;
        if (cn < H) {
            const mbf16* pb = W + (cn + col) * ldw + koff;
            #pragma unroll
            for (int s = 0; s < KS; ++s) b1[s] = ld_frag(pb + 32 * s);
        }
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        #pragma unroll
        for (int s = 0; s < KS; ++s)
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[s], b0[s], acc,
                                                          0, 0, 0);
        float bv = bias ? (float)bias[c + col] : 0.f;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int row = (lane >> 4) * 4 + r;
            long gm = m0 + row;
            float v = acc[r] + bv;
            if (relu) v = v > 0.f ? v : 0.f;
            if (mask_act && gm < M
                && !((float)mask_act[gm * mask_ld + c + col] > 0.f))
                v = 0.f;
            mbf16 hv = (mbf16)v;
            if (lds_out) lds_out[row * MLP_LD + c + col] = hv;
            if (gm < M) save[gm * save_ld + c + col] = hv;
        }
        #pragma unroll
        for (int s = 0; s < KS; ++s) b0[s] = b1[s];
    }
}

static __device__ __forceinline__ void mlp_layer(
        const mbf16* A, long lda, long m0, long M,
        const mbf16* W, long ldw, const mbf16* bias, long H, long Kp,
        const mbf16* mask_act, long mask_ld,
        mbf16* lds_out, mbf16* save, long save_ld,
        int wave, int lane, bool relu) {
    if (Kp == 256)
        return mlp_layer_u<8>(A, lda, m0, M, W, ldw, bias, H, mask_act,
                              mask_ld, lds_out, save, save_ld, wave, lane,
                              relu);
    if (Kp == 416)
        return mlp_layer_u<13>(A, lda, m0, M, W, ldw, bias, H, mask_act,
                               mask_ld, lds_out, save, save_ld, wave, lane,
                               relu);
    // generic (unpipelined) fallback for other shapes
    for (long c = wave * 16; c < H; c += 16 * MLP_WAVES) {
        f32x4 acc = tile16(A, lda, W, ldw, c, Kp, lane);
        const int col = lane & 15;
        float bv = bias ? (float)bias[c + col] : 0.f;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int row = (lane >> 4) * 4 + r;
            long gm = m0 + row;
            float v = acc[r] + bv;
            if (relu) v = v > 0.f ? v : 0.f;
            if (mask_act && gm < M
                && !((float)mask_act[gm * mask_ld + c + col] > 0.f))
                v = 0.f;
            mbf16 hv = (mbf16)v;
            if (lds_out) lds_out[row * MLP_LD + c + col] = hv;
            if (gm < M) save[gm * save_ld + c + col] = hv;
        }
    }
}

extern "C" __global__ __launch_bounds__(64 * MLP_WAVES, 2)
void k_mlp3_fwd(const mbf16* __restrict__ x0, long M, long K0p,
                const mbf16* __restrict__ w1,   // [H, K0p] padded
                const mbf16* __restrict__ b1,
                const mbf16* __restrict__ w2,   // [H, Hp] padded
                const mbf16* __restrict__ b2,
                const mbf16* __restrict__ w3,   // [H, Hp] padded
                const mbf16* __restrict__ b3,
                const mbf16* __restrict__ w4,   // [H]
                const mbf16* __restrict__ b4,
                const float* __restrict__ partial,  // [M] logit carry-in or null
                long H, long Hp,
                mbf16* __restrict__ a1, mbf16* __restrict__ a2,
                mbf16* __restrict__ a3, float* __restrict__ out) {
    __shared__ mbf16 act[2][MLP_BM * MLP_LD];
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const long m0 = (long)blockIdx.x * MLP_BM;
    if (m0 >= M) return;
    // LDS starts with arbitrary bits; the pad columns [H, MLP_LD) are read
    // by the next stage's fragments against ZERO weight pads — but
    // 0 * inf-garbage = NaN, so they must be zeroed (found the hard way:
    // NaN from step ~3 once other kernels had dirtied the LDS)
    for (int i = threadIdx.x; i < 2 * MLP_BM * MLP_LD; i += blockDim.x)
        act[0][i] = (mbf16)0.0f;
    __syncthreads();

    mlp_layer(x0 + m0 * K0p, K0p, m0, M, w1, K0p, b1, H, K0p,
              nullptr, 0, act[0], a1, H, wave, lane, true);
    __syncthreads();
    mlp_layer(act[0], MLP_LD, m0, M, w2, Hp, b2, H, Hp,
              nullptr, 0, act[1], a2, H, wave, lane, true);
    __syncthreads();
    mlp_layer(act[1], MLP_LD, m0, M, w3, Hp, b3, H, Hp,
              nullptr, 0, act[0], a3, H, wave, lane, true);
    __syncthreads();

    // final Linear(H, 1): VALU dot per row
    const int rows_per_wave = (MLP_BM + MLP_WAVES - 1) / MLP_WAVES;
    for (int r = 0; r < rows_per_wave; ++r) {
        long row = (long)wave * rows_per_wave + r;
        long gm = m0 + row;
        if (row >= MLP_BM || gm >= M) continue;
        float s = 0.f;
        for (long k = lane; k < H; k += 64)
            s += (float)act[0][row * MLP_LD + k] * (float)w4[k];
        #pragma unroll
        for (int off = 32; off; off >>= 1)
            s += __shfl_down(s, off, 64);
        if (lane == 0)
            out[gm] = s + (float)b4[0] + (partial ? partial[gm] : 0.f);
    }
}

extern "C" __global__ __launch_bounds__(64 * MLP_WAVES, 2)
void k_mlp3_bwd(const float* __restrict__ dout, long M, long K0p,
                const mbf16* __restrict__ a1, const mbf16* __restrict__ a2,
                const mbf16* __restrict__ a3,
                const mbf16* __restrict__ w4,    // [H]
                const mbf16* __restrict__ w3t,   // [H, Hp] padded transposed
                const mbf16* __restrict__ w2t,   // [H, Hp]
                const mbf16* __restrict__ w1t,   // [K0p, Hp]
                long H, long Hp,
                mbf16* __restrict__ dz1, mbf16* __restrict__ dz2,
                mbf16* __restrict__ dz3, mbf16* __restrict__ dx0,
                float* __restrict__ bias_out) {  // optional [4H+1] scratch
    __shared__ mbf16 dz[2][MLP_BM * MLP_LD];
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const long m0 = (long)blockIdx.x * MLP_BM;
    if (m0 >= M) return;
    for (int i = threadIdx.x; i < 2 * MLP_BM * MLP_LD; i += blockDim.x)
        dz[0][i] = (mbf16)0.0f;   // see fwd: 0 * LDS-garbage-inf = NaN
    __syncthreads();

    // dz3 = dout ⊗ w4 ⊙ relu'(a3), elementwise; zero the LDS pad columns
    // once (the B-side pads are zero too, but dz tiles are the A side of
    // the NEXT stage whose pads multiply W pads — either side zero is
    // enough; zeroing here keeps the invariant simple)
    for (long i = threadIdx.x; i < MLP_BM * MLP_LD; i += blockDim.x) {
        long row = i / MLP_LD, n = i % MLP_LD;
        long gm = m0 + row;
        float v = 0.f;
        if (n < H && gm < M && (float)a3[gm * H + n] > 0.f)
            v = dout[gm] * (float)w4[n];
        mbf16 hv = (mbf16)v;
        dz[0][i] = hv;
        if (n < H && gm < M) dz3[gm * H + n] = hv;
    }
    if (bias_out) {
        // head wgrad dw4[n] = sum_m dout[m]*a3[m,n] and db4 ride along as
        // a SECOND walk over the a3 tile the loop above just pulled into
        // L1/L2 (16 rows x H x 2B = 13 KB — hot): the separate head pass
        // over a3 (3 MB/step HBM) disappears, while the main loop keeps
        // its linear-index ILP (a fused column-walk version cost ~4 us)
        for (long n = threadIdx.x; n < H; n += blockDim.x) {
            float sw = 0.f, s4 = 0.f;
            for (int row = 0; row < MLP_BM; ++row) {
                long gm = m0 + row;
                if (gm < M) {
                    float dv = dout[gm];
                    sw += dv * (float)a3[gm * H + n];
                    if (n == 0) s4 += dv;
                }
            }
            atomicAdd(&bias_out[3 * H + n], sw);
            if (n == 0 && s4 != 0.f) atomicAdd(&bias_out[4 * H], s4);
        }
    }
    __syncthreads();
    // NOTE: dz tiles' pad columns [H, Hp) may hold garbage after a GEMM
    // stage — harmless, because the B side (padded weight copies) is zero
    // there, so pad products vanish.
    mlp_layer(dz[0], MLP_LD, m0, M, w3t, Hp, nullptr, H, Hp,
              a2, H, dz[1], dz2, H, wave, lane, false);
    __syncthreads();
    mlp_layer(dz[1], MLP_LD, m0, M, w2t, Hp, nullptr, H, Hp,
              a1, H, dz[0], dz1, H, wave, lane, false);
    __syncthreads();
    mlp_layer(dz[0], MLP_LD, m0, M, w1t, Hp, nullptr, K0p, Hp,
              nullptr, 0, nullptr, dx0, K0p, wave, lane, false);
}

// Bias grads for all 3 hidden layers + the head scalar in one pass over
// the dz mirrors k_mlp3_bwd already writes (torch did them as 3 GEMV
// launches + a reduce + an add, ~35 us/step at B=4096 H=400). Partials
// accumulate into a persistent fp32 scratch [3H+1] via atomics; the
// finisher below folds the scratch into the (bf16) grad buffers and
// re-zeros it, so the scratch is zero again before the next step's
// launch — the whole pair is hipGraph-capturable with no per-step fill.
extern "C" __global__ void k_mlp3_bias_bwd(
        const float* __restrict__ dout,
        const mbf16* __restrict__ dz1, const mbf16* __restrict__ dz2,
        const mbf16* __restrict__ dz3, const mbf16* __restrict__ a3,
        long M, long H, long rows_per_blk,
        float* __restrict__ scratch, int with_dz, int with_head) {
    const long r0 = (long)blockIdx.x * rows_per_blk;
    if (r0 >= M) return;
    const long r1 = min(M, r0 + rows_per_blk);
    // column sums: thread t covers columns t, t+blockDim.x, ... of each dz
    // (+ the head wgrad dw4[c] = sum_r dout[r]*a3[r,c] — same access shape).
    // with_dz == 0: the fused wgrad kernel already accumulated the three
    // dz column sums from its LDS-staged tiles; with_head == 0: the dgrad
    // kernel carried dw4/db4 (it reads dout/a3 anyway). Both 0: the
    // launcher never starts this kernel, only the finisher.
    for (long c = threadIdx.x; c < H; c += blockDim.x) {
        float sw = 0.f;
        if (with_dz) {
            float s1 = 0.f, s2 = 0.f, s3 = 0.f;
            for (long r = r0; r < r1; ++r) {
                s1 += (float)dz1[r * H + c];
                s2 += (float)dz2[r * H + c];
                s3 += (float)dz3[r * H + c];
                if (with_head) sw += dout[r] * (float)a3[r * H + c];
            }
            atomicAdd(scratch + c, s1);
            atomicAdd(scratch + H + c, s2);
            atomicAdd(scratch + 2 * H + c, s3);
        } else if (with_head) {
            for (long r = r0; r < r1; ++r)
                sw += dout[r] * (float)a3[r * H + c];
        }
        if (with_head) atomicAdd(scratch + 3 * H + c, sw);
    }
    if (with_head) {
        // head bias: sum of dout rows, one atomic per wave
        float s4 = 0.f;
        for (long r = r0 + threadIdx.x; r < r1; r += blockDim.x)
            s4 += dout[r];
        #pragma unroll
        for (int off = 32; off; off >>= 1) s4 += __shfl_down(s4, off, 64);
        if ((threadIdx.x & 63) == 0 && s4 != 0.f)
            atomicAdd(scratch + 4 * H, s4);
    }
}

extern "C" __global__ void k_mlp3_bias_finish(
        float* __restrict__ scratch, long H,
        mbf16* __restrict__ db1, mbf16* __restrict__ db2,
        mbf16* __restrict__ db3, mbf16* __restrict__ dw4,
        mbf16* __restrict__ db4) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i > 4 * H) return;
    float v = scratch[i];
    scratch[i] = 0.f;               // self-cleaning for the next step
    mbf16* dst = i < H        ? db1 + i
               : i < 2 * H    ? db2 + (i - H)
               : i < 3 * H    ? db3 + (i - 2 * H)
               : i < 4 * H    ? dw4 + (i - 3 * H)
               :                db4;
    *dst = (mbf16)((float)*dst + v);    // zero_grad zeroed it; += matches
}                                       // torch's beta=1 accumulation

extern "C" void emb_mlp3_bias_bwd(const float* dout, const void* dz1,
                                  const void* dz2, const void* dz3,
                                  const void* a3,
                                  long M, long H, float* scratch,
                                  void* db1, void* db2, void* db3,
                                  void* dw4, void* db4, int with_dz,
                                  int with_head, hipStream_t stream) {
    if (M == 0) return;
    const long rows_per_blk = 32;
    int ga = (int)((M + rows_per_blk - 1) / rows_per_blk);
    if (with_dz || with_head)
        k_mlp3_bias_bwd<<<ga, 256, 0, stream>>>(
            dout, (const mbf16*)dz1, (const mbf16*)dz2, (const mbf16*)dz3,
            (const mbf16*)a3, M, H, rows_per_blk, scratch, with_dz,
            with_head);
    int gb = (int)((4 * H + 1 + 255) / 256);
    k_mlp3_bias_finish<<<gb, 256, 0, stream>>>(
        scratch, H, (mbf16*)db1, (mbf16*)db2, (mbf16*)db3, (mbf16*)dw4,
        (mbf16*)db4);
}

extern "C" void emb_mlp3_fwd(const void* x0, long M, long K0p,
                             const void* w1, const void* b1,
                             const void* w2, const void* b2,
                             const void* w3, const void* b3,
                             const void* w4, const void* b4,
                             const float* partial,
                             long H, long Hp,
                             void* a1, void* a2, void* a3, float* out,
                             hipStream_t stream) {
    if (M == 0) return;
    long grid = (M + MLP_BM - 1) / MLP_BM;
    k_mlp3_fwd<<<(int)grid, 64 * MLP_WAVES, 0, stream>>>(
        (const mbf16*)x0, M, K0p, (const mbf16*)w1, (const mbf16*)b1,
        (const mbf16*)w2, (const mbf16*)b2, (const mbf16*)w3,
        (const mbf16*)b3, (const mbf16*)w4, (const mbf16*)b4, partial, H, Hp,
        (mbf16*)a1, (mbf16*)a2, (mbf16*)a3, out);
}

extern "C" void emb_mlp3_bwd(const float* dout, long M, long K0p,
                             const void* a1, const void* a2, const void* a3,
                             const void* w4, const void* w3t,
                             const void* w2t, const void* w1t,
                             long H, long Hp,
                             void* dz1, void* dz2, void* dz3, void* dx0,
                             float* bias_out, hipStream_t stream) {
    if (M == 0) return;
    long grid = (M + MLP_BM - 1) / MLP_BM;
    k_mlp3_bwd<<<(int)grid, 64 * MLP_WAVES, 0, stream>>>(
        dout, M, K0p, (const mbf16*)a1, (const mbf16*)a2, (const mbf16*)a3,
        (const mbf16*)w4, (const mbf16*)w3t, (const mbf16*)w2t,
        (const mbf16*)w1t, H, Hp, (mbf16*)dz1, (mbf16*)dz2, (mbf16*)dz3,
        (mbf16*)dx0, bias_out);
}

// ---------------------------------------------------------- fused wgrads
// dW_L[i,j] += sum_m dz_L[m,i] * a_{L-1}[m,j] for the 3 hidden layers in
// ONE launch. hipBLASLt ran these [H x M]@[M x H'] shapes at ~64 TF
// (64x16 macro tiles, 182 workgroups): here a block computes a 64x64
// fp32 tile with both operands LDS-transposed on load, M split 4 ways
// into an fp32 scratch that a finisher folds into the bf16 grads
// (+=, self-cleaning — the same capture-safe pattern as the bias pass).

#define WG_KC 128           // m-chunk staged per LDS round
#define WG_LD (WG_KC + 8)   // LDS row stride (16 B-aligned frag rows)

extern "C" __global__ __launch_bounds__(512, 2)
void k_mlp3_wgrad(const mbf16* __restrict__ dz1,   // [M, H]
                  const mbf16* __restrict__ dz2,   // [M, H]
                  const mbf16* __restrict__ dz3,   // [M, H]
                  const mbf16* __restrict__ x0,    // [M, K0p]
                  const mbf16* __restrict__ a1,    // [M, H]
                  const mbf16* __restrict__ a2,    // [M, H]
                  long M, long H, long K0p,
                  long t1, long tk, long th,       // L1 tiles, K0p/H tiles
                  float* __restrict__ scratch,     // [H*K0p + 2*H*H]
                  long m_split,
                  float* __restrict__ bias_out) {  // optional [3*H] bias
    // block -> (layer, i-tile, j-tile)
    long bt = blockIdx.x;
    const mbf16 *dz, *a;
    float* out;
    long ti, tj, jdim, bseg;
    if (bt < t1) {                    // L1: dz1^T @ x0 -> [H, K0p]
        dz = dz1; a = x0; out = scratch; jdim = K0p; bseg = 0;
        ti = bt / tk; tj = bt - ti * tk;
    } else if (bt < t1 + th * th) {   // L2: dz2^T @ a1 -> [H, H]
        bt -= t1;
        dz = dz2; a = a1; out = scratch + H * K0p; jdim = H; bseg = H;
        ti = bt / th; tj = bt - ti * th;
    } else {                          // L3: dz3^T @ a2 -> [H, H]
        bt -= t1 + th * th;
        dz = dz3; a = a2; out = scratch + H * K0p + H * H; jdim = H;
        bseg = 2 * H;
        ti = bt / th; tj = bt - ti * th;
    }
    const long i0 = ti * 64, j0 = tj * 64;
    const long msz = ((M / 32 + m_split - 1) / m_split) * 32;
    const long mbeg = (long)blockIdx.y * msz;
    const long mend = (mbeg + msz < M) ? (mbeg + msz) : M;
    if (mbeg >= M) return;

    __shared__ mbf16 ldsA[64 * (WG_KC + 8)];   // dz^T tile  [i][m]
    __shared__ mbf16 ldsB[64 * (WG_KC + 8)];   // a^T tile   [j][m]
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;

    // wave w owns output subtiles {w, w+8} of the 4x4 16x16 grid
    f32x4 acc[2];
    acc[0] = f32x4{0.f, 0.f, 0.f, 0.f};
    acc[1] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (long m0 = mbeg; m0 < mend; m0 += WG_KC) {
        __syncthreads();
        // stage transposed: read rows m (64 consecutive columns
        // coalesced), write LDS [col][m] — 512 threads x 8 elements
        const long mlim = (mend - m0 < WG_KC) ? (mend - m0) : WG_KC;
        for (int e = (int)threadIdx.x; e < 64 * WG_KC;
             e += (int)blockDim.x) {
            int mm = e >> 6, cc = e & 63;
            mbf16 va = (mbf16)0.0f, vb = (mbf16)0.0f;
            if (mm < mlim) {
                long m = m0 + mm;
                if (i0 + cc < H) va = dz[m * H + i0 + cc];
                if (j0 + cc < jdim)
                    vb = a[m * jdim + j0 + cc];
            }
            ldsA[cc * WG_LD + mm] = va;
            ldsB[cc * WG_LD + mm] = vb;
        }
        __syncthreads();
        if (bias_out && tj == 0) {
            // bias grads ride along: the dz tile this block just staged IS
            // the operand k_mlp3_bias_bwd used to re-read from HBM — sum
            // its columns here instead (tj==0 blocks only: each (layer,
            // i-tile, m-split) contributes once). 8 threads per column,
            // tree-reduced within the 8-lane group; zeros were staged
            // beyond mlim/H so no bounds handling is needed in the sum.
            int cc = (int)threadIdx.x >> 3, sub = (int)threadIdx.x & 7;
            float s = 0.f;
            for (int mm = sub; mm < WG_KC; mm += 8)
                s += (float)ldsA[cc * WG_LD + mm];
            s += __shfl_down(s, 4);
            s += __shfl_down(s, 2);
            s += __shfl_down(s, 1);
            if (sub == 0 && i0 + cc < H)
                atomicAdd(&bias_out[bseg + i0 + cc], s);
        }
        const long koff = (lane >> 4) * 8;
        #pragma unroll
        for (int t = 0; t < 2; ++t) {
            int st = wave + t * 8;            // subtile id 0..15
            int si = (st >> 2) * 16, sj = (st & 3) * 16;
            const mbf16* pa = ldsA + (si + (lane & 15)) * WG_LD + koff;
            const mbf16* pb = ldsB + (sj + (lane & 15)) * WG_LD + koff;
            for (int ks = 0; ks < WG_KC; ks += 32) {
                bf16x8 av = ld_frag(pa + ks);
                bf16x8 bv = ld_frag(pb + ks);
                acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    av, bv, acc[t], 0, 0, 0);
            }
        }
    }
    #pragma unroll
    for (int t = 0; t < 2; ++t) {
        int st = wave + t * 8;
        int si = (st >> 2) * 16, sj = (st & 3) * 16;
        long i = i0 + si + (lane >> 4) * 4;   // C row = A's m-dim = i
        long j = j0 + sj + (lane & 15);
        if (j >= jdim) continue;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            if (i + r >= H) break;
            atomicAdd(&out[(i + r) * jdim + j], acc[t][r]);
        }
    }
}

extern "C" __global__ void k_mlp3_wgrad_finish(
        float* __restrict__ scratch, long H, long K0p, long K0,
        mbf16* __restrict__ dw1, mbf16* __restrict__ dw2,
        mbf16* __restrict__ dw3) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long n1 = H * K0p, n23 = H * H;
    if (i >= n1 + 2 * n23) return;
    float v = scratch[i];
    scratch[i] = 0.f;                    // self-cleaning for the next step
    mbf16* dst;
    if (i < n1) {
        long row = i / K0p, col = i - row * K0p;
        if (col >= K0) return;           // x0 pad columns: zero, unfolded
        dst = dw1 + row * K0 + col;      // the PARAM is [H, K0] unpadded
    } else if (i < n1 + n23) {
        dst = dw2 + (i - n1);
    } else {
        dst = dw3 + (i - n1 - n23);
    }
    *dst = (mbf16)((float)*dst + v);     // += matches torch beta=1 accum
}

extern "C" void emb_mlp3_wgrad(const void* dz1, const void* dz2,
                               const void* dz3, const void* x0,
                               const void* a1, const void* a2,
                               long M, long H, long K0p, long K0,
                               float* scratch,
                               void* dw1, void* dw2, void* dw3,
                               float* bias_out,
                               hipStream_t stream) {
    if (!M) return;
    const long m_split = 8;
    long th = (H + 63) / 64, tk = (K0p + 63) / 64;
    long t1 = th * tk;
    long tiles = t1 + 2 * th * th;
    dim3 grid((unsigned)tiles, (unsigned)m_split);
    k_mlp3_wgrad<<<grid, 512, 0, stream>>>(
        (const mbf16*)dz1, (const mbf16*)dz2, (const mbf16*)dz3,
        (const mbf16*)x0, (const mbf16*)a1, (const mbf16*)a2, M, H, K0p,
        t1, tk, th, scratch, m_split, bias_out);
    long total = H * K0p + 2 * H * H;
    k_mlp3_wgrad_finish<<<(int)((total + 255) / 256), 256, 0, stream>>>(
        scratch, H, K0p, K0, (mbf16*)dw1, (mbf16*)dw2, (mbf16*)dw3);
}

// ---- fused weight repack ----------------------------------------------
// The train step refreshes six padded weight copies per iteration (three
// plain pads before mlp3_fwd, three transposed pads before mlp3_bwd); as
// separate aten::copy_ launches they cost ~3.8 us EACH while moving <1 MB
// (profiles/step_attrib_r2.txt) — pure launch/ramp. One kernel packs all
// three matrices of a phase. gridDim.y selects the matrix; the valid
// region only is written (padded tails were zeroed once at alloc and
// never change).

struct PackOne {
    const mbf16* src;   // [rows, src_ld] row-major, valid [rows, cols]
    mbf16* dst;         // plain: [rows, dst_ld]; trans: [cols, dst_ld]
    int rows, cols, src_ld, dst_ld, trans;
};

extern "C" __global__ void k_mlp3_pack(PackOne p0, PackOne p1, PackOne p2) {
    PackOne p = blockIdx.y == 0 ? p0 : (blockIdx.y == 1 ? p1 : p2);
    long n = (long)p.rows * p.cols;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (p.trans) {
        // iterate destination-linear: consecutive i -> consecutive r ->
        // coalesced writes; strided reads ride L2 (matrices are ~100s KB)
        int c = (int)(i / p.rows), r = (int)(i % p.rows);
        p.dst[(long)c * p.dst_ld + r] = p.src[(long)r * p.src_ld + c];
    } else {
        int r = (int)(i / p.cols), c = (int)(i % p.cols);
        p.dst[(long)r * p.dst_ld + c] = p.src[(long)r * p.src_ld + c];
    }
}

extern "C" void emb_mlp3_pack(const void* w1, void* d1, long r1, long c1,
                              long sld1, long dld1, int t1,
                              const void* w2, void* d2, long r2, long c2,
                              long sld2, long dld2, int t2,
                              const void* w3, void* d3, long r3, long c3,
                              long sld3, long dld3, int t3,
                              hipStream_t stream) {
    PackOne p0{(const mbf16*)w1, (mbf16*)d1, (int)r1, (int)c1, (int)sld1,
               (int)dld1, t1};
    PackOne p1{(const mbf16*)w2, (mbf16*)d2, (int)r2, (int)c2, (int)sld2,
               (int)dld2, t2};
    PackOne p2{(const mbf16*)w3, (mbf16*)d3, (int)r3, (int)c3, (int)sld3,
               (int)dld3, t3};
    long mx = r1 * c1;
    if (r2 * c2 > mx) mx = r2 * c2;
    if (r3 * c3 > mx) mx = r3 * c3;
    if (!mx) return;
    dim3 grid((unsigned)((mx + 255) / 256), 3);
    k_mlp3_pack<<<grid, 256, 0, stream>>>(p0, p1, p2);
}
