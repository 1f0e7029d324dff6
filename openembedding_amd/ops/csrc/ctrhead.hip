// Fused CTR interaction head (gfx950 / MI355X).
//
// The DeepFM/WDL "glue" between the embedding gather and the MLP GEMMs was
// ~320 us/step of small elementwise/reduce kernels + ~110 us of autocast
// casts (profiles/bench_deepfm_1gpu_kernels.md). These two kernels fuse, in
// one pass over the gathered rows:
//
//   forward:  e_all [B, F, D1] (D1 = dim+1: dim embedding cols + the merged
//             first-order/"wide" column — see models/ctr.py) and dense
//             [B, ND] ->
//               deep_in  bf16 [B, F*dim + ND]   (cast fused into the write)
//               partial  f32  [B] = sum_f e[.,dim]            (first order)
//                              + 0.5*sum_d((sum_f e)^2 - sum_f e^2)  (FM)
//                              + dense @ w + b             (dense linear)
//   backward: d_deep_in bf16, d_partial f32 ->
//               de_all f32 (FM + first-order + deep contributions),
//               d_dense f32, dw/db accumulated with atomics.
//
// One 64-lane wave per sample; lane c covers column c of the row (D1 <= 64
// enforced host-side), so the per-field row load is one coalesced segment.
// Reference math: DeepFM second order 0.5*((sum e)^2 - sum e^2) — the same
// formula the TF/DeepCTR models of the reference benchmark compute
// (test/benchmark/criteo_deepctr.py DeepFM).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef long long i64;
typedef unsigned long long u64;
typedef __hip_bfloat16 bf16;

// lanes c < dim: embedding columns; lane dim: wide column; F fields.
// use_fm: 0 for WDL/LR-style heads.
template <typename OutT>
__global__ void k_ctr_head_fwd(const float* __restrict__ e_all,
                               const float* __restrict__ dense,
                               const float* __restrict__ w,  // [nd]
                               const float* __restrict__ bias,  // [1]
                               long B, long F, long dim, long nd,
                               long out_stride,
                               OutT* __restrict__ deep_in,
                               float* __restrict__ partial,
                               float* __restrict__ s_out,  // [B, dim] for bwd
                               int use_fm) {
    const long D1 = dim + 1;
    const int lane = threadIdx.x & 63;
    const long b = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    if (b >= B) return;
    const float* e = e_all + b * F * D1;
    OutT* di = deep_in + b * out_stride;
    // zero the alignment pad tail (the fused MLP reads 16-byte fragments)
    for (long j = F * dim + nd + lane; j < out_stride; j += 64)
        di[j] = (OutT)0.0f;

    // lane covers columns lane and lane+64 (D1 <= 128 — covers the
    // reference's dim-64 benchmark config, D1 = 65)
    float s[2] = {0.0f, 0.0f}, sq[2] = {0.0f, 0.0f};
    for (long f = 0; f < F; ++f) {
        #pragma unroll
        for (int t = 0; t < 2; ++t) {
            const long cc = lane + 64 * t;
            if (cc >= D1) break;
            float v = e[f * D1 + cc];
            if (cc < dim) {
                s[t] += v;
                sq[t] += v * v;
                di[f * dim + cc] = (OutT)v;
            } else {
                s[t] += v;                // first-order (wide) column
            }
        }
    }
    // dense tail: cast + dot
    float dsum = 0.0f;
    for (long j = lane; j < nd; j += 64) {
        float v = dense[b * nd + j];
        di[F * dim + j] = (OutT)v;
        dsum += v * w[j];
    }
    float fm = 0.0f, lin = 0.0f;
    #pragma unroll
    for (int t = 0; t < 2; ++t) {
        const long cc = lane + 64 * t;
        if (cc >= D1) break;
        if (cc < dim) {
            if (use_fm) fm += s[t] * s[t] - sq[t];
            if (s_out) s_out[b * dim + cc] = s[t];
        } else {
            lin = s[t];
        }
    }
    float acc = 0.5f * fm + lin + dsum;
    for (int off = 32; off; off >>= 1)
        acc += __shfl_down(acc, off, 64);
    if (lane == 0) partial[b] = acc + bias[0];
}

// backward, fully parallel: one thread per element of de_all [B, F, D1]
// (the wave-per-sample version serialized F iterations over ~D1 active
// lanes and measured 69 us/step; with s saved by the forward this is a pure
// elementwise pass at memory speed).
template <typename OutT>
__global__ void k_ctr_head_bwd_e(const float* __restrict__ e_all,
                                 const OutT* __restrict__ d_deep_in,
                                 const float* __restrict__ d_partial,
                                 const float* __restrict__ s_in, // [B, dim]
                                 long B, long F, long dim,
                                 long nd, long out_stride,
                                 float* __restrict__ de_all,
                                 int use_fm,
                                 float* __restrict__ dw,
                                 float* __restrict__ db) {
    const long D1 = dim + 1;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    // zero the tiny dense-linear grad accumulators consumed by the
    // bwd_d kernel that follows on this stream (two torch::zeros
    // allocations were ~4 us of launch each for ~56 bytes)
    if (i < nd) dw[i] = 0.0f;
    if (i == nd) *db = 0.0f;
    if (i >= B * F * D1) return;
    const long c = i % D1;
    const long f = (i / D1) % F;
    const long b = i / (D1 * F);
    const float gp = d_partial[b];
    if (c == dim) { de_all[i] = gp; return; }        // wide column
    float g = (float)d_deep_in[b * out_stride + f * dim + c];
    if (use_fm) g += gp * (s_in[b * dim + c] - e_all[i]);
    de_all[i] = g;
}

// dense tail, one thread per (sample, feature): coalesced d_dense/dense/
// d_deep_in traffic, dw/db via LDS block partials then nd+1 global atomics
// per block (the thread-per-sample version had only B/256 blocks in flight
// and measured 34.6 us; this shape is ~population-bound).
#define ND_MAX 32
template <typename OutT>
__global__ void k_ctr_head_bwd_d(const float* __restrict__ dense,
                                 const float* __restrict__ w,
                                 const OutT* __restrict__ d_deep_in,
                                 const float* __restrict__ d_partial,
                                 long B, long F, long dim, long nd,
                                 long out_stride,
                                 float* __restrict__ d_dense,
                                 float* __restrict__ dw,   // [nd] atomic
                                 float* __restrict__ db) { // [1] atomic
    __shared__ float lacc[ND_MAX + 1];
    for (int t = threadIdx.x; t < nd + 1; t += blockDim.x) lacc[t] = 0.0f;
    __syncthreads();
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < B * nd) {
        const long b = i / nd;
        const long j = i % nd;
        const float gp = d_partial[b];
        const float dv = dense[i];
        d_dense[i] = (float)d_deep_in[b * out_stride + F * dim + j]
                     + gp * w[j];
        atomicAdd(&lacc[j], gp * dv);
        if (j == 0) atomicAdd(&lacc[nd], gp);
    }
    __syncthreads();
    for (int t = threadIdx.x; t < nd + 1; t += blockDim.x) {
        if (t < nd) atomicAdd(&dw[t], lacc[t]);
        else atomicAdd(db, lacc[nd]);
    }
}

extern "C" {

void emb_ctr_head_fwd(const float* e_all, const float* dense, const float* w,
                      const float* bias, long B, long F, long dim, long nd,
                      long out_stride, void* deep_in, float* partial,
                      float* s_out, int use_fm, int out_bf16,
                      hipStream_t stream) {
    if (B == 0) return;
    int block = 256;                    // 4 waves per block
    long grid = (B * 64 + block - 1) / block;
    if (out_bf16)
        k_ctr_head_fwd<bf16><<<(int)grid, block, 0, stream>>>(
            e_all, dense, w, bias, B, F, dim, nd, out_stride,
            (bf16*)deep_in, partial, s_out, use_fm);
    else
        k_ctr_head_fwd<float><<<(int)grid, block, 0, stream>>>(
            e_all, dense, w, bias, B, F, dim, nd, out_stride,
            (float*)deep_in, partial, s_out, use_fm);
}

void emb_ctr_head_bwd(const float* e_all, const float* dense, const float* w,
                      const void* d_deep_in, const float* d_partial,
                      const float* s_in,
                      long B, long F, long dim, long nd, long out_stride,
                      float* de_all, float* d_dense, float* dw, float* db,
                      int use_fm, int out_bf16, hipStream_t stream) {
    if (B == 0) return;
    int block = 256;
    long grid_e = (B * F * (dim + 1) + block - 1) / block;
    long grid_d = (B * nd + block - 1) / block;
    if (out_bf16) {
        k_ctr_head_bwd_e<bf16><<<(int)grid_e, block, 0, stream>>>(
            e_all, (const bf16*)d_deep_in, d_partial, s_in, B, F, dim, nd,
            out_stride, de_all, use_fm, dw, db);
        k_ctr_head_bwd_d<bf16><<<(int)grid_d, block, 0, stream>>>(
            dense, w, (const bf16*)d_deep_in, d_partial, B, F, dim, nd,
            out_stride, d_dense, dw, db);
    } else {
        k_ctr_head_bwd_e<float><<<(int)grid_e, block, 0, stream>>>(
            e_all, (const float*)d_deep_in, d_partial, s_in, B, F, dim, nd,
            out_stride, de_all, use_fm, dw, db);
        k_ctr_head_bwd_d<float><<<(int)grid_d, block, 0, stream>>>(
            dense, w, (const float*)d_deep_in, d_partial, B, F, dim, nd,
            out_stride, d_dense, dw, db);
    }
}

}  // extern "C"
