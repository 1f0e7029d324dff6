// CIN implicit-GEMM kernels (gfx950, bf16 MFMA 16x16x32).
//
// One CIN layer is  out[n, o] = sum_{f,h} W[o, f*H+h] * x0[n, f] * xk[n, h]
// over columns n = dd*B + b (d-leading layout, matching models/ctr.py's
// _CINLayerFn). The operand V[n, k] = x0[n, k/H] * xk[n, k%H] is an outer
// product per column — torch paths must materialize it (245 MB bf16 per
// build at the benchmark shape; the round-1 einsum made it 490 MB fp32).
// Here V is built on the fly in LDS per 128-column block, so the only
// HBM traffic is x0/xk/W/out — measured profile said the V builds +
// re-reads were ~1.2 ms of the 1.67 ms xDeepFM step.
//
// Fragment maps are the ones verified by mlp.hip/tests:
//   A[m][k]: m = lane&15, k = (lane>>4)*8 + e   (A = W[o][k] row-major)
//   B[k][n]: n = lane&15, k = (lane>>4)*8 + e   (B read from V[n][k] rows)
//   C/D:     col = lane&15, row = (lane>>4)*4 + r
//
// Geometry: block = 128 columns x all O outputs; 8 waves; K streamed in
// chunks of CIN_KC built into LDS (bf16, padded row stride breaks the
// 16-lane fragment-read bank cycle). Requires O % 16 == 0, O <= 128,
// F <= 32, H <= 128 (the zoo shapes; ctr.py falls back to the chunked
// torch path otherwise).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __hip_bfloat16 cbf16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define CIN_COLS 128          // columns (b,dd pairs) per block
#define CIN_KC 128            // K chunk built per LDS round (4 MFMA steps)
#define CIN_VLD (CIN_KC + 8)  // LDS row stride: 68 dwords, 16-lane clean
#define CIN_WAVES 8

static __device__ __forceinline__ bf16x8 cin_ld_frag(const cbf16* p) {
    return *reinterpret_cast<const bf16x8*>(__builtin_assume_aligned(p, 16));
}

extern "C" __global__ __launch_bounds__(64 * CIN_WAVES, 2)
void k_cin_fwd(const float* __restrict__ x0p,   // [N, F]  (d-leading cols)
               const float* __restrict__ xkp,   // [N, H]
               const cbf16* __restrict__ w,     // [O, Kp] padded bf16
               float* __restrict__ out,         // [N, O]
               long N, long F, long H, long O, long Kp) {
    __shared__ cbf16 vtile[CIN_COLS * CIN_VLD];
    __shared__ cbf16 x0c[CIN_COLS * 32];          // F <= 32, bf16 staged
    __shared__ cbf16 xkc[CIN_COLS * 128];         // H <= 128
    __shared__ short ftab[CIN_KC], htab[CIN_KC];  // k -> (f, h) per chunk
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const long n0 = (long)blockIdx.x * CIN_COLS;
    const int K = (int)(F * H);
    const int Fi = (int)F, Hi = (int)H;

    // All per-element index math is INT32 (64-bit div/mod emulation cost
    // ~10x here), operands staged as bf16 (pre-rounded once — the torch
    // path rounds identically), and the V build writes PACKED bf16x8: the
    // first version was ISSUE-bound on the scalar build (PMC: 35% active
    // issue, MFMA a small fraction) at 1 block/CU; this shape runs 2
    // blocks/CU with ~3x fewer issue slots per element.

    for (int base = 0; base < CIN_COLS * Fi; base += (int)blockDim.x) {
        int i = base + (int)threadIdx.x;
        if (i >= CIN_COLS * Fi) break;
        int c = i / Fi, f = i - c * Fi;
        x0c[c * 32 + f] = (cbf16)((n0 + c < N) ? x0p[(n0 + c) * F + f]
                                               : 0.f);
    }
    for (int base = 0; base < CIN_COLS * Hi; base += (int)blockDim.x) {
        int i = base + (int)threadIdx.x;
        if (i >= CIN_COLS * Hi) break;
        int c = i / Hi, h = i - c * Hi;
        xkc[c * 128 + h] = (cbf16)((n0 + c < N) ? xkp[(n0 + c) * H + h]
                                                : 0.f);
    }

    // wave owns o rows [wave*16, wave*16+16) x all 8 column subtiles
    const long o0 = (long)wave * 16;
    f32x4 acc[8];
    #pragma unroll
    for (int t = 0; t < 8; ++t) acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

    // first chunk's W fragment issued before any barrier; later chunks'
    // first fragments are issued during the previous chunk's MFMA phase,
    // so the W stream's L2 latency hides behind build + barriers
    const long koff0 = (lane >> 4) * 8;
    bf16x8 a_pre;
    if (o0 < O)
        a_pre = cin_ld_frag(w + (o0 + (lane & 15)) * Kp + koff0);
    for (long k0 = 0; k0 < Kp; k0 += CIN_KC) {
        __syncthreads();
        if (threadIdx.x < CIN_KC) {
            int k = (int)k0 + (int)threadIdx.x;
            int f = (k < K) ? k / Hi : 0;
            ftab[threadIdx.x] = (short)f;
            htab[threadIdx.x] = (short)(k - f * Hi);
        }
        __syncthreads();
        // packed build: 8 V elements -> one 16 B LDS store
        const int k0i = (int)k0;
        for (int g = (int)threadIdx.x; g < CIN_COLS * (CIN_KC / 8);
             g += (int)blockDim.x) {
            int c = g >> 4;               // CIN_KC/8 == 16 groups per row
            int kko = (g & 15) << 3;
            bf16x8 vv;
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                int kk = kko + j;
                float v = 0.f;
                if (k0i + kk < K)
                    v = (float)x0c[c * 32 + ftab[kk]]
                        * (float)xkc[c * 128 + htab[kk]];
                vv[j] = (__bf16)v;
            }
            *reinterpret_cast<bf16x8*>(vtile + c * CIN_VLD + kko) = vv;
        }
        __syncthreads();
        if (o0 >= O) continue;      // narrow O: spare waves still build V
        const long kc_lim = (Kp - k0 < CIN_KC) ? (Kp - k0) : CIN_KC;
        const cbf16* pa = w + (o0 + (lane & 15)) * Kp + k0 + koff0;
        const cbf16* pb = vtile + (lane & 15) * CIN_VLD + koff0;
        bf16x8 a0 = a_pre;
        for (long ks = 0; ks < kc_lim; ks += 32) {
            bf16x8 a1;
            if (ks + 32 < kc_lim) a1 = cin_ld_frag(pa + ks + 32);
            else if (k0 + CIN_KC < Kp)      // next chunk's first fragment
                a1 = cin_ld_frag(pa + CIN_KC);
            #pragma unroll
            for (int t = 0; t < 8; ++t) {
                bf16x8 b = cin_ld_frag(pb + t * 16 * CIN_VLD + ks);
                acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a0, b, acc[t], 0, 0, 0);
            }
            a0 = a1;
        }
        a_pre = a0;                         // carries the prefetched frag
    }

    // epilogue: each lane holds 4 consecutive o values per subtile
    if (o0 < O) {
        #pragma unroll
        for (int t = 0; t < 8; ++t) {
            long n = n0 + t * 16 + (lane & 15);
            if (n >= N) continue;
            long ob = o0 + (lane >> 4) * 4;
            float* dst = out + n * O + ob;
            #pragma unroll
            for (int r = 0; r < 4; ++r) dst[r] = acc[t][r];
        }
    }
}

extern "C" void emb_cin_fwd(const float* x0p, const float* xkp,
                            const void* w, float* out, long N, long F,
                            long H, long O, long Kp, hipStream_t stream) {
    if (!N) return;
    long grid = (N + CIN_COLS - 1) / CIN_COLS;
    k_cin_fwd<<<(int)grid, 64 * CIN_WAVES, 0, stream>>>(
        x0p, xkp, (const cbf16*)w, out, N, F, H, O, Kp);
}

// ------------------------------------------------------------ weight grad
// dW[o, f*H+h] = sum_n dZ[n,o] x0[n,f] xk[n,h]. Reframed per f as a GEMM
// over n with the B operand built on the fly: B'[n][h] = xkt[h][n] *
// x0t[f][n]. Grid = (f, n-split); per block: all [O x H] tiles accumulate
// over the block's n-range in registers, then one fp32 atomicAdd pass into
// dW (split-K). Operands dZt [O, Np], xkt [H, Np], x0t [F, Np] are bf16
// transposed copies (library transposes, ~20 MB total); Np % 32 == 0 with
// zero tails.

extern "C" __global__ __launch_bounds__(64 * CIN_WAVES, 1)
void k_cin_dw(const cbf16* __restrict__ dzt,   // [O, Np]
              const cbf16* __restrict__ x0t,   // [F, Np]
              const cbf16* __restrict__ xkt,   // [H, Np]
              float* __restrict__ dw,          // [O, F*H] fp32 accum
              long Np, long F, long H, long O, long n_split) {
    __shared__ cbf16 btile[128 * 72];   // [h][64 n], stride 72 (16B rows)
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const long f = blockIdx.x;
    const long split = blockIdx.y;
    const long nsz = ((Np / 32 + n_split - 1) / n_split) * 32;
    const long nbeg = split * nsz;
    const long nend = (nbeg + nsz < Np) ? (nbeg + nsz) : Np;
    if (nbeg >= Np) return;
    const long o0 = (long)wave * 16;

    f32x4 acc[8];
    #pragma unroll
    for (int t = 0; t < 8; ++t) acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

    const long koff = (lane >> 4) * 8;
    const int Hp16 = (int)((H + 15) / 16 * 16);  // partial h-tile zeroed
    const int Hi = (int)H;
    const cbf16* pa = dzt + (o0 + (lane & 15)) * Np + koff;
    // A-stream pipelined across 64-wide n-chunks (2 MFMA k-steps each):
    // the next step's dZt fragment loads behind the build + barrier
    bf16x8 a_cur;
    if (o0 < O) a_cur = cin_ld_frag(pa + nbeg);
    for (long n0s = nbeg; n0s < nend; n0s += 64) {
        __syncthreads();
        // B' chunk: [H][64] = xkt rows * x0 row f (broadcast over h),
        // packed: two bf16x8 loads -> one bf16x8 store per 8 elements
        const long nrem = nend - n0s;
        for (int g = (int)threadIdx.x; g < Hp16 * 8;
             g += (int)blockDim.x) {
            int h = g >> 3, no = (g & 7) << 3;
            bf16x8 vv;
            if (h < Hi && no < nrem) {
                bf16x8 xk8 = cin_ld_frag(xkt + h * Np + n0s + no);
                bf16x8 x08 = cin_ld_frag(x0t + f * Np + n0s + no);
                #pragma unroll
                for (int j = 0; j < 8; ++j)
                    vv[j] = (__bf16)((float)xk8[j] * (float)x08[j]);
            } else {
                #pragma unroll
                for (int j = 0; j < 8; ++j) vv[j] = (__bf16)0.f;
            }
            *reinterpret_cast<bf16x8*>(btile + h * 72 + no) = vv;
        }
        __syncthreads();
        if (o0 >= O) continue;
        const long ks_lim = (nrem < 64) ? nrem : 64;
        for (long ks = 0; ks < ks_lim; ks += 32) {
            bf16x8 a = a_cur;
            if (n0s + ks + 32 < nend)
                a_cur = cin_ld_frag(pa + n0s + ks + 32);
            #pragma unroll
            for (int t = 0; t < 8; ++t) {
                if (t * 16 >= Hi) break;
                bf16x8 b = cin_ld_frag(btile + (t * 16 + (lane & 15)) * 72
                                       + koff + ks);
                acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a, b, acc[t], 0, 0, 0);
            }
        }
    }
    if (o0 >= O) return;
    #pragma unroll
    for (int t = 0; t < 8; ++t) {
        long h = t * 16 + (lane & 15);
        if (h >= H) continue;   // partial tile: never cross into f+1's h=0
        long ob = o0 + (lane >> 4) * 4;
        #pragma unroll
        for (int r = 0; r < 4; ++r)
            atomicAdd(&dw[(ob + r) * (F * H) + f * H + h], acc[t][r]);
    }
}

extern "C" void emb_cin_dw(const void* dzt, const void* x0t, const void* xkt,
                           float* dw, long Np, long F, long H, long O,
                           long n_split, hipStream_t stream) {
    if (!Np) return;
    dim3 grid((unsigned)F, (unsigned)n_split);
    k_cin_dw<<<grid, 64 * CIN_WAVES, 0, stream>>>(
        (const cbf16*)dzt, (const cbf16*)x0t, (const cbf16*)xkt, dw, Np, F,
        H, O, n_split);
}

// --------------------------------------------------------- input gradients
// Per column n:  P[k] = sum_o W[o,k] dZ[n,o]   (GEMM over O, K-tiled)
//   dx0[n,f] = sum_h P[f*H+h] xk[n,h]
//   dxk[n,h] = sum_f P[f*H+h] x0[n,f]
// P never touches HBM: K is walked one FIELD at a time (k in [f*H,(f+1)*H)
// is exactly h), each wave computes whole k-tiles of the chunk's P into an
// LDS staging tile, and the consume phase is plain bank-padded LDS
// arithmetic — NO atomics (the first version accumulated through LDS
// float atomics and measured 1.2 ms/call: contention turns them into
// serialized CAS traffic). Wt [Kp, Op] is the bf16 transposed weight copy;
// block = 64 columns.

#define CDX_COLS 64
#define CDX_X0LD 36     // x0/dx0 LDS row stride (bank-cycling pad)
#define CDX_XKLD 132    // xk/dxk LDS row stride
#define CDX_PLD 72      // pchunk row stride

extern "C" __global__ __launch_bounds__(64 * CIN_WAVES, 1)
void k_cin_dx(const float* __restrict__ doutp,  // [N, O] fp32
              const cbf16* __restrict__ wt,     // [Kp, Op]
              const float* __restrict__ x0p,    // [N, F]
              const float* __restrict__ xkp,    // [N, H]
              float* __restrict__ dx0p,         // [N, F] out
              float* __restrict__ dxkp,         // [N, H] out
              long N, long F, long H, long O, long Op) {
    __shared__ cbf16 dzc[CDX_COLS * 136];        // [c][o], stride 136
    __shared__ float x0c[CDX_COLS * CDX_X0LD];   // [c][f]
    __shared__ float xkc[CDX_COLS * CDX_XKLD];   // [c][h]
    __shared__ float dx0a[CDX_COLS * CDX_X0LD];  // [c][f] accum
    __shared__ float dxka[CDX_COLS * CDX_XKLD];  // [c][h] accum
    __shared__ float pchunk[128 * CDX_PLD];      // [h][c] P staging
    __shared__ float dotred[8 * 72];             // dx0 dot partials
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const long n0 = (long)blockIdx.x * CDX_COLS;
    const int Fi = (int)F, Hi = (int)H;

    for (int i = (int)threadIdx.x; i < CDX_COLS * CDX_X0LD;
         i += (int)blockDim.x)
        dx0a[i] = 0.f;
    for (int i = (int)threadIdx.x; i < CDX_COLS * CDX_XKLD;
         i += (int)blockDim.x)
        dxka[i] = 0.f;
    for (int i = (int)threadIdx.x; i < CDX_COLS * Fi;
         i += (int)blockDim.x) {
        int c = i / Fi, ff = i - (i / Fi) * Fi;
        x0c[c * CDX_X0LD + ff] = (n0 + c < N) ? x0p[(n0 + c) * F + ff]
                                              : 0.f;
    }
    for (int i = (int)threadIdx.x; i < CDX_COLS * Hi;
         i += (int)blockDim.x) {
        int c = i / Hi, h = i - (i / Hi) * Hi;
        xkc[c * CDX_XKLD + h] = (n0 + c < N) ? xkp[(n0 + c) * H + h]
                                             : 0.f;
    }
    for (int i = (int)threadIdx.x; i < CDX_COLS * 128;
         i += (int)blockDim.x) {
        int c = i >> 7, o = i & 127;
        float v = (o < O && n0 + c < N) ? doutp[(n0 + c) * O + o] : 0.f;
        dzc[c * 136 + o] = (cbf16)v;
    }

    const long koff = (lane >> 4) * 8;
    const int htiles = (Hi + 15) / 16;
    // Wt fragment prefetch: the NEXT field's loads are issued before the
    // consume barrier, so their L2 latency hides behind the consume phase
    bf16x8 a_pre[4];
    const bool owns_tile = wave < htiles;   // one k-tile per wave per field
    if (Op == 128 && owns_tile) {
        const cbf16* pa0 = wt + ((long)wave * 16 + (lane & 15)) * Op + koff;
        #pragma unroll
        for (int j = 0; j < 4; ++j) a_pre[j] = cin_ld_frag(pa0 + 32 * j);
    }
    for (int f = 0; f < Fi; ++f) {
        __syncthreads();   // previous consume done before P overwrite
        // this field's P [H x cols]: wave w owns k-tiles w, w+8, ...
        for (int kt = wave; kt < htiles; kt += CIN_WAVES) {
            const long k0 = (long)f * Hi + kt * 16;
            const cbf16* pa = wt + (k0 + (lane & 15)) * Op + koff;
            bf16x8 a[4];
            if (Op == 128) {
                if (kt == wave) {
                    #pragma unroll
                    for (int j = 0; j < 4; ++j) a[j] = a_pre[j];
                } else {
                    #pragma unroll
                    for (int j = 0; j < 4; ++j)
                        a[j] = cin_ld_frag(pa + 32 * j);
                }
            }
            #pragma unroll
            for (int cs = 0; cs < CDX_COLS / 16; ++cs) {
                f32x4 acc = {0.f, 0.f, 0.f, 0.f};
                const cbf16* pb = dzc + (cs * 16 + (lane & 15)) * 136
                                  + koff;
                if (Op == 128) {
                    #pragma unroll
                    for (int j = 0; j < 4; ++j) {
                        bf16x8 b = cin_ld_frag(pb + 32 * j);
                        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a[j], b, acc, 0, 0, 0);
                    }
                } else {
                    for (long oc = 0; oc < Op; oc += 32) {
                        bf16x8 av = cin_ld_frag(pa + oc);
                        bf16x8 b = cin_ld_frag(pb + oc);
                        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            av, b, acc, 0, 0, 0);
                    }
                }
                // P subtile store: lane rows h = kt*16 + (lane>>4)*4 + r
                int c = cs * 16 + (lane & 15);
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int h = kt * 16 + (lane >> 4) * 4 + r;
                    if (h < Hi) pchunk[h * CDX_PLD + c] = acc[r];
                }
            }
        }
        if (Op == 128 && owns_tile && f + 1 < Fi) {
            const cbf16* pan = wt + ((long)(f + 1) * Hi + wave * 16
                                     + (lane & 15)) * Op + koff;
            #pragma unroll
            for (int j = 0; j < 4; ++j) a_pre[j] = cin_ld_frag(pan + 32 * j);
        }
        __syncthreads();
        // consume — fused, balanced, no atomics: thread (hr, c) walks
        // h = hr, hr+8, ..., updating dxk elementwise and accumulating a
        // dx0 partial; partials reduce across the 8 h-groups through LDS
        {
            int c = (int)threadIdx.x & 63;
            int hr = (int)threadIdx.x >> 6;
            float x0f = x0c[c * CDX_X0LD + f];
            float s = 0.f;
            for (int h = hr; h < Hi; h += 8) {
                float p = pchunk[h * CDX_PLD + c];
                dxka[c * CDX_XKLD + h] += p * x0f;
                s += p * xkc[c * CDX_XKLD + h];
            }
            dotred[hr * 72 + c] = s;
        }
        __syncthreads();
        if (threadIdx.x < CDX_COLS) {
            int c = (int)threadIdx.x;
            float s = 0.f;
            #pragma unroll
            for (int hr = 0; hr < 8; ++hr) s += dotred[hr * 72 + c];
            dx0a[c * CDX_X0LD + f] = s;
        }
    }
    __syncthreads();
    for (int i = (int)threadIdx.x; i < CDX_COLS * Fi;
         i += (int)blockDim.x) {
        int c = i / Fi, ff = i - (i / Fi) * Fi;
        if (n0 + c < N) dx0p[(n0 + c) * F + ff] = dx0a[c * CDX_X0LD + ff];
    }
    for (int i = (int)threadIdx.x; i < CDX_COLS * Hi;
         i += (int)blockDim.x) {
        int c = i / Hi, h = i - (i / Hi) * Hi;
        if (n0 + c < N) dxkp[(n0 + c) * H + h] = dxka[c * CDX_XKLD + h];
    }
}


extern "C" void emb_cin_dx(const float* doutp, const void* wt,
                           const float* x0p, const float* xkp, float* dx0p,
                           float* dxkp, long N, long F, long H, long O,
                           long Op, hipStream_t stream) {
    if (!N) return;
    long grid = (N + CDX_COLS - 1) / CDX_COLS;
    k_cin_dx<<<(int)grid, 64 * CIN_WAVES, 0, stream>>>(
        doutp, (const cbf16*)wt, x0p, xkp, dx0p, dxkp, N, F, H, O, Op);
}
