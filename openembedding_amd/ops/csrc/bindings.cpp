// Torch bindings for the openembedding_amd CDNA4 kernels (embops.hip).
// Tensor-level API consumed by ops/dispatch.py and core/variable_gpu.py.
//
// Two calling modes:
//   exact   — unique_inverse() syncs once to return a tightly-sized unique
//             tensor (needed for the multi-GPU all_to_all splits anyway);
//   bounded — unique_bounded() returns an n-sized buffer + device count and
//             every downstream kernel takes the count as a device pointer,
//             so a whole train step runs with ZERO host syncs (and is
//             hipGraph-capturable).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>  // ROCm torch masquerades HIP as CUDA here
#include <c10/cuda/CUDAGuard.h>

#include <cstdint>
#include <tuple>
#include <vector>

// torch instantiates data_ptr<int64_t> (long on LP64); the kernel TU uses
// long long — same width, C linkage, so the pointer casts below are safe.
typedef unsigned long long u64;
typedef int64_t i64;
typedef struct ihipStream_t* hipStream_t_;

extern "C" {
void emb_unique(const i64*, long, u64*, int*, long, int*, unsigned char*,
                i64*, i64*, int*, const i64*, int, hipStream_t_);
void emb_ht_lookup(u64*, int*, long, const i64*, long, int*, i64*, i64*,
                   unsigned char*, int, const int*, long, hipStream_t_);
void emb_ht_rehash(const u64*, const int*, long, u64*, int*, long,
                   hipStream_t_);
void emb_array_touch(unsigned char*, const i64*, long, long, long, i64*,
                     unsigned char*, const int*, hipStream_t_);
void emb_gather_init(float*, float*, long, long, const i64*,
                     const unsigned char*, const i64*, long, const i64*,
                     float*, int, float, float, float, u64, const float*,
                     const int*, hipStream_t_);
void emb_reduce_by_inverse(const i64*, const float*, long, long, float*, u64*,
                           long, hipStream_t_);
void emb_apply_optimizer(int, float*, float*, long, long, const i64*, long,
                         const float*, const u64*, const float*, const int*,
                         hipStream_t_);
void emb_ctr_head_fwd(const float*, const float*, const float*, const float*, long,
                      long, long, long, long, void*, float*, float*, int,
                      int, hipStream_t_);
void emb_ctr_head_bwd(const float*, const float*, const float*, const void*,
                      const float*, const float*, long, long, long, long,
                      long, float*, float*, float*, float*, int, int,
                      hipStream_t_);
void emb_mlp3_bias_bwd(const float*, const void*, const void*, const void*,
                       const void*, long, long, float*, void*, void*, void*,
                       void*, void*, int, int, hipStream_t_);
void emb_mlp3_wgrad(const void*, const void*, const void*, const void*,
                    const void*, const void*, long, long, long, long,
                    float*, void*, void*, void*, float*, hipStream_t_);
void emb_mlp3_pack(const void*, void*, long, long, long, long, int,
                   const void*, void*, long, long, long, long, int,
                   const void*, void*, long, long, long, long, int,
                   hipStream_t_);
void emb_bce_fwd(const float*, const float*, long, float*, hipStream_t_);
void emb_bce_bwd(const float*, const float*, long, const float*, float*,
                 hipStream_t_);
void emb_bucketize_pad(const i64*, long, const int*, long, long, i64*, int*,
                       int*, int*, int*, hipStream_t_);
void emb_mask_tail(i64*, float*, u64*, long, long, const int*, hipStream_t_);
void emb_fault_in(const i64*, long, const int*, const u64*, const int*, long,
                  const float*, const float*, float*, float*, long, long,
                  const i64*, unsigned char*, int*, hipStream_t_);
void emb_spill_rows(const i64*, const i64*, long, const float*, const float*,
                    float*, float*, long, long, hipStream_t_);
void emb_gather_host(const i64*, long, const i64*, const u64*, const int*,
                     long, const float*, float*, long, hipStream_t_);
void emb_gather_pad(const float*, const u64*, long, const int*, long, float*,
                    hipStream_t_);
void emb_scatter_out(const float*, const i64*, long, const int*, long, float*,
                     hipStream_t_);
void emb_split_payload(const float*, long, long, const int*, float*, u64*,
                       hipStream_t_);
void emb_flat_step_scalars(float*, float, float, hipStream_t_);
void emb_flat_opt(int, void*, float*, float*, float*, void*,
                  const float*, long, int, float, float, float, float,
                  hipStream_t_);
void emb_flat_adagrad_f32(float*, float*, const float*, long, float, float,
                          hipStream_t_);
void emb_flat_adagrad_bf16(float*, float*, const void*, void*, long, float,
                           float, hipStream_t_);
void emb_cin_fwd(const float*, const float*, const void*, float*, long,
                 long, long, long, long, hipStream_t_);
void emb_cin_dw(const void*, const void*, const void*, float*, long, long,
                long, long, long, hipStream_t_);
void emb_cin_dx(const float*, const void*, const float*, const float*,
                float*, float*, long, long, long, long, long, hipStream_t_);
void emb_mlp3_fwd(const void*, long, long, const void*, const void*,
                  const void*, const void*, const void*, const void*,
                  const void*, const void*, const float*, long, long, void*,
                  void*, void*, float*, hipStream_t_);
void emb_mlp3_bwd(const float*, long, long, const void*, const void*,
                  const void*, const void*, const void*, const void*,
                  const void*, long, long, void*, void*, void*, void*,
                  float*, hipStream_t_);
}

namespace {

using OptTensor = c10::optional<torch::Tensor>;

hipStream_t_ cur_stream() {
    return reinterpret_cast<hipStream_t_>(
        at::cuda::getCurrentCUDAStream().stream());
}

long next_pow2(long x) {
    long p = 16;
    while (p < x) p <<= 1;
    return p;
}

const int* u_ptr(const OptTensor& u_dev) {
    if (!u_dev.has_value()) return nullptr;
    TORCH_CHECK(u_dev->dtype() == torch::kInt32, "u_dev must be int32");
    return u_dev->data_ptr<int>();
}

#define CHECK_GPU(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")
#define CHECK_CONT(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

// ---- unique ----------------------------------------------------------

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> unique_bounded(
    torch::Tensor keys, OptTensor field_offsets) {
    CHECK_GPU(keys); CHECK_CONT(keys);
    TORCH_CHECK(keys.dtype() == torch::kInt64);
    const i64* foff = nullptr;
    int F = 0;
    if (field_offsets.has_value() && field_offsets->numel()) {
        CHECK_GPU(*field_offsets); CHECK_CONT(*field_offsets);
        TORCH_CHECK(field_offsets->dtype() == torch::kInt64,
                    "field_offsets must be int64");
        TORCH_CHECK(keys.numel() % field_offsets->numel() == 0,
                    "keys length must be a multiple of field_offsets");
        foff = field_offsets->data_ptr<i64>();
        F = (int)field_offsets->numel();
    }
    const c10::cuda::CUDAGuard guard(keys.device());
    long n = keys.numel();
    auto opts_i64 = keys.options();
    auto opts_i32 = keys.options().dtype(torch::kInt32);
    auto opts_u8 = keys.options().dtype(torch::kUInt8);
    long cap = next_pow2(2 * n);
    auto tk = torch::empty({cap}, opts_i64);
    auto tv = torch::empty({cap}, opts_i32);
    auto slot_of = torch::empty({n}, opts_i32);
    auto is_first = torch::empty({n}, opts_u8);
    auto uk = torch::empty({n}, opts_i64);  // tail unread
    auto inverse = torch::empty({n}, opts_i64);
    auto counter = torch::empty({1}, opts_i32);
    emb_unique(keys.data_ptr<i64>(), n, (u64*)tk.data_ptr<i64>(),
               tv.data_ptr<int>(), cap, slot_of.data_ptr<int>(),
               is_first.data_ptr<uint8_t>(), uk.data_ptr<i64>(),
               inverse.data_ptr<i64>(), counter.data_ptr<int>(), foff, F,
               cur_stream());
    return {uk, inverse, counter};
}

std::tuple<torch::Tensor, torch::Tensor> unique_inverse(torch::Tensor keys) {
    auto r = unique_bounded(keys, c10::nullopt);
    long u = std::get<2>(r).item<int>();  // host sync
    return {std::get<0>(r).narrow(0, 0, u), std::get<1>(r)};
}

// ---- persistent hash table -------------------------------------------

std::tuple<torch::Tensor, torch::Tensor> ht_lookup(
    torch::Tensor tk, torch::Tensor tv, torch::Tensor keys,
    torch::Tensor nrows, torch::Tensor slot_keys, bool insert,
    OptTensor u_dev) {
    CHECK_GPU(keys); CHECK_CONT(keys);
    const c10::cuda::CUDAGuard guard(keys.device());
    long n = keys.numel();
    auto slots = torch::empty({n}, keys.options());
    auto new_mask = torch::empty({n}, keys.options().dtype(torch::kUInt8));
    if (n)
        emb_ht_lookup((u64*)tk.data_ptr<i64>(), tv.data_ptr<int>(),
                      tk.numel(), keys.data_ptr<i64>(), n,
                      nrows.data_ptr<int>(), slot_keys.data_ptr<i64>(),
                      slots.data_ptr<i64>(), new_mask.data_ptr<uint8_t>(),
                      insert ? 1 : 0, u_ptr(u_dev), slot_keys.numel(),
                      cur_stream());
    return {slots, new_mask};
}

void ht_rehash(torch::Tensor tk_old, torch::Tensor tv_old, torch::Tensor tk_new,
               torch::Tensor tv_new) {
    const c10::cuda::CUDAGuard guard(tk_old.device());
    emb_ht_rehash((u64*)tk_old.data_ptr<i64>(), tv_old.data_ptr<int>(),
                  tk_old.numel(), (u64*)tk_new.data_ptr<i64>(),
                  tv_new.data_ptr<int>(), tk_new.numel(), cur_stream());
}

// ---- array table ------------------------------------------------------

std::tuple<torch::Tensor, torch::Tensor> array_touch(
    torch::Tensor valid, torch::Tensor keys, int64_t shard_num,
    OptTensor u_dev) {
    const c10::cuda::CUDAGuard guard(valid.device());
    long n = keys.numel();
    auto slots = torch::empty({n}, keys.options());
    auto new_mask = torch::empty({n}, keys.options().dtype(torch::kUInt8));
    if (n)
        emb_array_touch(valid.data_ptr<uint8_t>(), keys.data_ptr<i64>(), n,
                        shard_num, valid.numel(), slots.data_ptr<i64>(),
                        new_mask.data_ptr<uint8_t>(), u_ptr(u_dev),
                        cur_stream());
    return {slots, new_mask};
}

// ---- gather + lazy init ----------------------------------------------

torch::Tensor gather_init(torch::Tensor weights, torch::Tensor state,
                          torch::Tensor slots, torch::Tensor new_mask,
                          torch::Tensor keys, OptTensor inverse,
                          int64_t init_cat, double p0, double p1, double p2,
                          int64_t seed, torch::Tensor state_init_row,
                          bool want_out, OptTensor u_dev) {
    CHECK_GPU(weights); CHECK_CONT(weights);
    const c10::cuda::CUDAGuard guard(weights.device());
    long n = inverse.has_value() ? inverse->numel() : keys.numel();
    long dim = weights.size(1);
    long sd = state.numel() ? state.size(1) : 0;
    torch::Tensor out;
    float* out_ptr = nullptr;
    if (want_out) {
        out = torch::empty({n, dim}, weights.options());
        out_ptr = out.data_ptr<float>();
    } else {
        out = torch::empty({0}, weights.options());
    }
    emb_gather_init(weights.data_ptr<float>(),
                    sd ? state.data_ptr<float>() : nullptr, dim, sd,
                    slots.data_ptr<i64>(),
                    new_mask.numel() ? new_mask.data_ptr<uint8_t>() : nullptr,
                    keys.data_ptr<i64>(), n,
                    inverse.has_value() ? inverse->data_ptr<i64>() : nullptr,
                    out_ptr, (int)init_cat,
                    (float)p0, (float)p1, (float)p2, (u64)seed,
                    sd ? state_init_row.data_ptr<float>() : nullptr,
                    u_ptr(u_dev), cur_stream());
    return out;
}

// ---- reduce-by-key ----------------------------------------------------

std::tuple<torch::Tensor, torch::Tensor> reduce_by_inverse(
    torch::Tensor inverse, torch::Tensor grads, int64_t u) {
    CHECK_GPU(grads); CHECK_CONT(grads);
    TORCH_CHECK(grads.dtype() == torch::kFloat32,
                "reduce_by_inverse: grads must be float32");
    const c10::cuda::CUDAGuard guard(grads.device());
    long n = grads.size(0);
    long dim = grads.size(1);
    auto ugrads = torch::empty({u, dim}, grads.options());
    auto counts = torch::empty({u}, grads.options().dtype(torch::kInt64));
    emb_reduce_by_inverse(inverse.data_ptr<i64>(), grads.data_ptr<float>(), n,
                          dim, ugrads.data_ptr<float>(),
                          (u64*)counts.data_ptr<i64>(), u, cur_stream());
    return {ugrads, counts};
}

void mask_tail(torch::Tensor keys, torch::Tensor grads, torch::Tensor counts,
               OptTensor u_dev) {
    CHECK_GPU(keys); CHECK_CONT(keys); CHECK_CONT(grads); CHECK_CONT(counts);
    TORCH_CHECK(grads.size(0) == keys.numel()
                && counts.numel() == keys.numel(), "mask_tail sizes");
    const c10::cuda::CUDAGuard guard(keys.device());
    emb_mask_tail(keys.data_ptr<i64>(), grads.data_ptr<float>(),
                  (u64*)counts.data_ptr<i64>(), keys.numel(), grads.size(1),
                  u_ptr(u_dev), cur_stream());
}

// ---- capacity tier v2 --------------------------------------------------
// Pinned host tensors are device-accessible on ROCm (hipHostMalloc-backed
// via the torch pinned allocator): the fault-in/spill kernels read/write
// them directly — only touched rows cross the host link.

#define CHECK_PINNED(t) \
    TORCH_CHECK((t).is_pinned(), #t " must be a pinned host tensor")

void fault_in(torch::Tensor keys, OptTensor u_dev, torch::Tensor htk,
              torch::Tensor htv, torch::Tensor host_w, OptTensor host_s,
              torch::Tensor weights, torch::Tensor state, torch::Tensor slots,
              torch::Tensor new_mask, torch::Tensor faulted) {
    CHECK_GPU(keys); CHECK_CONT(keys); CHECK_CONT(host_w);
    CHECK_PINNED(host_w);
    const c10::cuda::CUDAGuard guard(keys.device());
    long sd = state.numel() ? state.size(1) : 0;
    const float* hs = nullptr;
    if (sd) {
        TORCH_CHECK(host_s.has_value(), "state tier needs host_s");
        CHECK_PINNED(*host_s);
        hs = host_s->data_ptr<float>();
    }
    emb_fault_in(keys.data_ptr<i64>(), keys.numel(), u_ptr(u_dev),
                 (const u64*)htk.data_ptr<i64>(), htv.data_ptr<int>(),
                 htk.numel(), host_w.data_ptr<float>(), hs,
                 weights.data_ptr<float>(),
                 sd ? state.data_ptr<float>() : nullptr, weights.size(1), sd,
                 slots.data_ptr<i64>(), new_mask.data_ptr<uint8_t>(),
                 faulted.data_ptr<int>(), cur_stream());
}

void spill_rows(torch::Tensor cache_slots, torch::Tensor host_slots,
                torch::Tensor weights, torch::Tensor state,
                torch::Tensor host_w, OptTensor host_s) {
    CHECK_GPU(cache_slots); CHECK_CONT(cache_slots); CHECK_CONT(host_slots);
    CHECK_PINNED(host_w);
    const c10::cuda::CUDAGuard guard(cache_slots.device());
    long sd = state.numel() ? state.size(1) : 0;
    float* hs = nullptr;
    if (sd) {
        TORCH_CHECK(host_s.has_value(), "state tier needs host_s");
        CHECK_PINNED(*host_s);
        hs = host_s->data_ptr<float>();
    }
    emb_spill_rows(cache_slots.data_ptr<i64>(), host_slots.data_ptr<i64>(),
                   cache_slots.numel(), weights.data_ptr<float>(),
                   sd ? state.data_ptr<float>() : nullptr,
                   host_w.data_ptr<float>(), hs, weights.size(1), sd,
                   cur_stream());
}

void gather_host(torch::Tensor keys, torch::Tensor cache_slots,
                 torch::Tensor htk, torch::Tensor htv, torch::Tensor host_w,
                 torch::Tensor out) {
    CHECK_GPU(keys); CHECK_CONT(keys); CHECK_CONT(out);
    CHECK_PINNED(host_w);
    const c10::cuda::CUDAGuard guard(keys.device());
    emb_gather_host(keys.data_ptr<i64>(), keys.numel(),
                    cache_slots.data_ptr<i64>(),
                    (const u64*)htk.data_ptr<i64>(), htv.data_ptr<int>(),
                    htk.numel(), host_w.data_ptr<float>(),
                    out.data_ptr<float>(), out.size(1), cur_stream());
}

// ---- padded all-to-all -------------------------------------------------

void bucketize_pad(torch::Tensor uk_buf, OptTensor u_dev, int64_t world,
                   int64_t cap, torch::Tensor send_keys,
                   torch::Tensor send_src, torch::Tensor pos_of,
                   torch::Tensor counts, torch::Tensor overflow) {
    CHECK_GPU(uk_buf); CHECK_CONT(uk_buf); CHECK_CONT(send_keys);
    CHECK_CONT(send_src); CHECK_CONT(pos_of);
    TORCH_CHECK(send_keys.numel() == world * cap &&
                send_src.numel() == world * cap &&
                pos_of.numel() >= uk_buf.numel() &&
                counts.numel() == world && overflow.numel() == 1,
                "bucketize_pad buffer sizes");
    TORCH_CHECK(send_src.dtype() == torch::kInt32 &&
                pos_of.dtype() == torch::kInt32 &&
                counts.dtype() == torch::kInt32 &&
                overflow.dtype() == torch::kInt32, "int32 buffers expected");
    const c10::cuda::CUDAGuard guard(uk_buf.device());
    emb_bucketize_pad(uk_buf.data_ptr<i64>(), uk_buf.numel(), u_ptr(u_dev),
                      world, cap, send_keys.data_ptr<i64>(),
                      send_src.data_ptr<int>(), pos_of.data_ptr<int>(),
                      counts.data_ptr<int>(), overflow.data_ptr<int>(),
                      cur_stream());
}

torch::Tensor gather_pad(torch::Tensor ugrads, torch::Tensor counts,
                         torch::Tensor send_src) {
    CHECK_GPU(ugrads); CHECK_CONT(ugrads); CHECK_CONT(counts);
    CHECK_CONT(send_src);
    TORCH_CHECK(ugrads.dtype() == torch::kFloat32 &&
                counts.dtype() == torch::kInt64 &&
                send_src.dtype() == torch::kInt32, "gather_pad dtypes");
    const c10::cuda::CUDAGuard guard(ugrads.device());
    long total = send_src.numel();
    long dim = ugrads.size(1);
    auto send_p = torch::empty({total, dim + 1}, ugrads.options());
    emb_gather_pad(ugrads.data_ptr<float>(), (const u64*)counts.data_ptr<i64>(),
                   dim, send_src.data_ptr<int>(), total,
                   send_p.data_ptr<float>(), cur_stream());
    return send_p;
}

torch::Tensor scatter_out(torch::Tensor rows_recv, torch::Tensor inverse,
                          torch::Tensor pos_of, int64_t n_elems) {
    CHECK_GPU(rows_recv); CHECK_CONT(rows_recv); CHECK_CONT(inverse);
    CHECK_CONT(pos_of);
    TORCH_CHECK(rows_recv.dtype() == torch::kFloat32 &&
                pos_of.dtype() == torch::kInt32, "scatter_out dtypes");
    const c10::cuda::CUDAGuard guard(rows_recv.device());
    long dim = rows_recv.size(1);
    auto out = torch::empty({n_elems, dim}, rows_recv.options());
    emb_scatter_out(rows_recv.data_ptr<float>(), inverse.data_ptr<i64>(),
                    n_elems, pos_of.data_ptr<int>(), dim,
                    out.data_ptr<float>(), cur_stream());
    return out;
}

std::tuple<torch::Tensor, torch::Tensor> split_payload(torch::Tensor g2c,
                                                       OptTensor u_dev) {
    CHECK_GPU(g2c); CHECK_CONT(g2c);
    TORCH_CHECK(g2c.dtype() == torch::kFloat32, "split_payload dtype");
    const c10::cuda::CUDAGuard guard(g2c.device());
    long u = g2c.size(0);
    long dim = g2c.size(1) - 1;
    // tail rows beyond *u_dev stay uninitialized; every consumer
    // (apply_optimizer) is u_dev-guarded and never reads them
    auto grads = torch::empty({u, dim}, g2c.options());
    auto counts = torch::empty({u}, g2c.options().dtype(torch::kInt64));
    emb_split_payload(g2c.data_ptr<float>(), u, dim, u_ptr(u_dev),
                      grads.data_ptr<float>(), (u64*)counts.data_ptr<i64>(),
                      cur_stream());
    return {grads, counts};
}

// ---- fused optimizers --------------------------------------------------

void apply_optimizer(int64_t opt, torch::Tensor weights, torch::Tensor state,
                     torch::Tensor slots, torch::Tensor grads,
                     torch::Tensor counts, std::vector<double> cfg,
                     OptTensor u_dev) {
    CHECK_GPU(weights); CHECK_CONT(weights); CHECK_CONT(grads);
    const c10::cuda::CUDAGuard guard(weights.device());
    long n = slots.numel();
    long dim = weights.size(1);
    long sd = state.numel() ? state.size(1) : 0;
    float c[7] = {0, 0, 0, 0, 0, 0, 0};
    for (size_t i = 0; i < cfg.size() && i < 7; ++i) c[i] = (float)cfg[i];
    emb_apply_optimizer((int)opt, weights.data_ptr<float>(),
                        sd ? state.data_ptr<float>() : nullptr, dim, sd,
                        slots.data_ptr<i64>(), n, grads.data_ptr<float>(),
                        (const u64*)counts.data_ptr<i64>(), c, u_ptr(u_dev),
                        cur_stream());
}

// ---- fused CTR interaction head ---------------------------------------

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> ctr_head_fwd(
    torch::Tensor e_all, torch::Tensor dense, torch::Tensor w,
    torch::Tensor bias, bool use_fm, bool out_bf16, bool pad32) {
    CHECK_GPU(e_all); CHECK_CONT(e_all); CHECK_CONT(dense); CHECK_CONT(w);
    const c10::cuda::CUDAGuard guard(e_all.device());
    long B = e_all.size(0), F = e_all.size(1), D1 = e_all.size(2);
    long dim = D1 - 1, nd = dense.size(1);
    TORCH_CHECK(D1 <= 128, "ctr_head: dim+1 must be <= 128");
    TORCH_CHECK(nd <= 32, "ctr_head: dense features must be <= 32");
    auto out_opts = e_all.options().dtype(out_bf16 ? torch::kBFloat16
                                                   : torch::kFloat32);
    long width = F * dim + nd;
    long stride = pad32 ? ((width + 31) / 32) * 32 : width;
    auto deep_in = torch::empty({B, stride}, out_opts);
    auto partial = torch::empty({B}, e_all.options());
    auto s_out = torch::empty({B, dim}, e_all.options());
    CHECK_CONT(bias);
    emb_ctr_head_fwd(e_all.data_ptr<float>(), dense.data_ptr<float>(),
                     w.data_ptr<float>(), bias.data_ptr<float>(), B, F, dim, nd,
                     stride, deep_in.data_ptr(), partial.data_ptr<float>(),
                     s_out.data_ptr<float>(),
                     use_fm ? 1 : 0, out_bf16 ? 1 : 0, cur_stream());
    return {deep_in, partial, s_out};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
ctr_head_bwd(torch::Tensor e_all, torch::Tensor dense, torch::Tensor w,
             torch::Tensor d_deep_in, torch::Tensor d_partial,
             torch::Tensor s_in, bool use_fm) {
    CHECK_GPU(e_all); CHECK_CONT(e_all); CHECK_CONT(d_deep_in);
    const c10::cuda::CUDAGuard guard(e_all.device());
    long B = e_all.size(0), F = e_all.size(1), D1 = e_all.size(2);
    long dim = D1 - 1, nd = dense.size(1);
    long stride = d_deep_in.size(1);  // may be 32-padded
    bool out_bf16 = d_deep_in.dtype() == torch::kBFloat16;
    auto de_all = torch::empty_like(e_all);
    auto d_dense = torch::empty_like(dense);
    // dw/db are zeroed by k_ctr_head_bwd_e before bwd_d accumulates
    auto dw = torch::empty_like(w);
    auto db = torch::empty({1}, w.options());
    if (B == 0) { dw.zero_(); db.zero_(); }
    emb_ctr_head_bwd(e_all.data_ptr<float>(), dense.data_ptr<float>(),
                     w.data_ptr<float>(), d_deep_in.data_ptr(),
                     d_partial.data_ptr<float>(),
                     s_in.data_ptr<float>(), B, F, dim, nd, stride,
                     de_all.data_ptr<float>(), d_dense.data_ptr<float>(),
                     dw.data_ptr<float>(), db.data_ptr<float>(),
                     use_fm ? 1 : 0, out_bf16 ? 1 : 0, cur_stream());
    return {de_all, d_dense, dw, db};
}

// ---- flat dense Adagrad -----------------------------------------------

void flat_adagrad(torch::Tensor param, torch::Tensor accum,
                  torch::Tensor grad, c10::optional<torch::Tensor> master,
                  double lr, double eps) {
    CHECK_GPU(param); CHECK_CONT(param); CHECK_CONT(accum); CHECK_CONT(grad);
    const c10::cuda::CUDAGuard guard(param.device());
    long n = param.numel();
    if (param.dtype() == torch::kFloat32) {
        emb_flat_adagrad_f32(param.data_ptr<float>(), accum.data_ptr<float>(),
                             grad.data_ptr<float>(), n, (float)lr, (float)eps,
                             cur_stream());
    } else {
        TORCH_CHECK(param.dtype() == torch::kBFloat16 && master.has_value(),
                    "bf16 flat_adagrad needs a float32 master tensor");
        emb_flat_adagrad_bf16(master->data_ptr<float>(),
                              accum.data_ptr<float>(), grad.data_ptr(),
                              param.data_ptr(), n, (float)lr, (float)eps,
                              cur_stream());
    }
}

void flat_step_scalars(torch::Tensor sc, double b1, double b2) {
    CHECK_GPU(sc); CHECK_CONT(sc);
    TORCH_CHECK(sc.numel() >= 3 && sc.dtype() == torch::kFloat32,
                "step scalars must be float32 [3]");
    const c10::cuda::CUDAGuard guard(sc.device());
    emb_flat_step_scalars(sc.data_ptr<float>(), (float)b1, (float)b2,
                          cur_stream());
}

void flat_opt(int64_t opt, torch::Tensor param, OptTensor master,
              OptTensor s1, OptTensor s2, torch::Tensor grad, OptTensor sc,
              double lr, double c0, double c1, double c2) {
    CHECK_GPU(param); CHECK_CONT(param); CHECK_CONT(grad);
    const c10::cuda::CUDAGuard guard(param.device());
    long n = param.numel();
    bool bf16 = param.dtype() == torch::kBFloat16;
    float* mp = nullptr;
    if (bf16) {
        TORCH_CHECK(master.has_value(), "bf16 flat_opt needs fp32 master");
        mp = master->data_ptr<float>();
    } else {
        TORCH_CHECK(param.dtype() == torch::kFloat32, "flat_opt dtype");
    }
    emb_flat_opt((int)opt, param.data_ptr(), mp,
                 s1.has_value() ? s1->data_ptr<float>() : nullptr,
                 s2.has_value() ? s2->data_ptr<float>() : nullptr,
                 grad.data_ptr(),
                 sc.has_value() ? sc->data_ptr<float>() : nullptr,
                 n, bf16 ? 1 : 0, (float)lr, (float)c0, (float)c1,
                 (float)c2, cur_stream());
}

// ---- fused MLP bias grads ---------------------------------------------

void mlp3_bias_bwd(torch::Tensor dout, torch::Tensor dz1, torch::Tensor dz2,
                   torch::Tensor dz3, torch::Tensor a3, torch::Tensor scratch,
                   torch::Tensor db1, torch::Tensor db2, torch::Tensor db3,
                   torch::Tensor dw4, torch::Tensor db4, bool with_dz,
                   bool with_head) {
    CHECK_GPU(dout); CHECK_CONT(dout); CHECK_CONT(dz1); CHECK_CONT(dz2);
    CHECK_CONT(dz3); CHECK_CONT(a3); CHECK_CONT(scratch);
    long M = dz1.size(0), H = dz1.size(1);
    TORCH_CHECK(dout.numel() == M && scratch.numel() >= 4 * H + 1 &&
                a3.sizes() == dz1.sizes(), "mlp3_bias_bwd shape mismatch");
    TORCH_CHECK(db1.dtype() == torch::kBFloat16 &&
                scratch.dtype() == torch::kFloat32,
                "mlp3_bias_bwd wants bf16 grads + fp32 scratch");
    TORCH_CHECK(db1.is_contiguous() && db2.is_contiguous() &&
                db3.is_contiguous() && dw4.is_contiguous() &&
                db4.is_contiguous() &&
                db1.numel() == H && db2.numel() == H && db3.numel() == H &&
                dw4.numel() == H && db4.numel() == 1, "bias grad layout");
    const c10::cuda::CUDAGuard guard(dout.device());
    emb_mlp3_bias_bwd(dout.data_ptr<float>(), dz1.data_ptr(), dz2.data_ptr(),
                      dz3.data_ptr(), a3.data_ptr(), M, H,
                      scratch.data_ptr<float>(),
                      db1.data_ptr(), db2.data_ptr(), db3.data_ptr(),
                      dw4.data_ptr(), db4.data_ptr(), with_dz ? 1 : 0,
                      with_head ? 1 : 0, cur_stream());
}

// ---- CIN implicit-GEMM -------------------------------------------------

torch::Tensor cin_fwd(torch::Tensor x0p, torch::Tensor xkp,
                      torch::Tensor wp) {
    CHECK_GPU(x0p); CHECK_CONT(x0p); CHECK_CONT(xkp); CHECK_CONT(wp);
    TORCH_CHECK(x0p.dtype() == torch::kFloat32
                && xkp.dtype() == torch::kFloat32
                && wp.dtype() == torch::kBFloat16, "cin_fwd dtypes");
    const c10::cuda::CUDAGuard guard(x0p.device());
    long N = x0p.size(0), F = x0p.size(1), H = xkp.size(1);
    long O = wp.size(0), Kp = wp.size(1);
    TORCH_CHECK(F <= 32 && H <= 128 && O <= 128 && O % 16 == 0
                && Kp % 32 == 0 && xkp.size(0) == N, "cin_fwd shapes");
    auto out = torch::empty({N, O}, x0p.options());
    emb_cin_fwd(x0p.data_ptr<float>(), xkp.data_ptr<float>(), wp.data_ptr(),
                out.data_ptr<float>(), N, F, H, O, Kp, cur_stream());
    return out;
}

torch::Tensor cin_dw(torch::Tensor dzt, torch::Tensor x0t,
                     torch::Tensor xkt, int64_t O, int64_t n_split) {
    CHECK_GPU(dzt); CHECK_CONT(dzt); CHECK_CONT(x0t); CHECK_CONT(xkt);
    TORCH_CHECK(dzt.dtype() == torch::kBFloat16
                && x0t.dtype() == torch::kBFloat16
                && xkt.dtype() == torch::kBFloat16, "cin_dw dtypes");
    const c10::cuda::CUDAGuard guard(dzt.device());
    long Np = dzt.size(1), F = x0t.size(0), H = xkt.size(0);
    TORCH_CHECK(Np % 32 == 0 && x0t.size(1) == Np && xkt.size(1) == Np
                && H <= 128 && O <= 128, "cin_dw shapes");
    auto dw = torch::zeros({O, F * H},
                           dzt.options().dtype(torch::kFloat32));
    emb_cin_dw(dzt.data_ptr(), x0t.data_ptr(), xkt.data_ptr(),
               dw.data_ptr<float>(), Np, F, H, O, n_split, cur_stream());
    return dw;
}

std::tuple<torch::Tensor, torch::Tensor> cin_dx(
        torch::Tensor doutp, torch::Tensor wt, torch::Tensor x0p,
        torch::Tensor xkp) {
    CHECK_GPU(doutp); CHECK_CONT(doutp); CHECK_CONT(wt); CHECK_CONT(x0p);
    CHECK_CONT(xkp);
    TORCH_CHECK(doutp.dtype() == torch::kFloat32
                && wt.dtype() == torch::kBFloat16, "cin_dx dtypes");
    const c10::cuda::CUDAGuard guard(doutp.device());
    long N = doutp.size(0), O = doutp.size(1);
    long F = x0p.size(1), H = xkp.size(1), Op = wt.size(1);
    TORCH_CHECK(Op % 32 == 0 && O <= Op && F <= 32 && H <= 128
                && wt.size(0) >= F * H, "cin_dx shapes");
    auto dx0 = torch::empty({N, F}, x0p.options());
    auto dxk = torch::empty({N, H}, xkp.options());
    emb_cin_dx(doutp.data_ptr<float>(), wt.data_ptr(),
               x0p.data_ptr<float>(), xkp.data_ptr<float>(),
               dx0.data_ptr<float>(), dxk.data_ptr<float>(), N, F, H, O,
               Op, cur_stream());
    return {dx0, dxk};
}

void mlp3_wgrad(torch::Tensor dz1, torch::Tensor dz2, torch::Tensor dz3,
                torch::Tensor x0, torch::Tensor a1, torch::Tensor a2,
                torch::Tensor scratch, torch::Tensor dw1, torch::Tensor dw2,
                torch::Tensor dw3, OptTensor bias_scratch) {
    // bias_scratch ([4H+1] fp32, the bias pass scratch): when given, the
    // kernel also accumulates the three dz column sums (bias grads) from
    // its LDS-staged tiles into segments 0/H/2H
    CHECK_GPU(dz1); CHECK_CONT(dz1); CHECK_CONT(dz2); CHECK_CONT(dz3);
    CHECK_CONT(x0); CHECK_CONT(a1); CHECK_CONT(a2); CHECK_CONT(scratch);
    TORCH_CHECK(dz1.dtype() == torch::kBFloat16
                && dw1.dtype() == torch::kBFloat16
                && scratch.dtype() == torch::kFloat32, "mlp3_wgrad dtypes");
    long M = dz1.size(0), H = dz1.size(1), K0p = x0.size(1);
    long K0 = dw1.size(1);
    TORCH_CHECK(M % 32 == 0, "mlp3_wgrad: M must be x32");
    TORCH_CHECK(scratch.numel() >= H * K0p + 2 * H * H
                && dw1.size(0) == H && K0 <= K0p && dw2.numel() == H * H
                && dw3.numel() == H * H
                && dw1.is_contiguous() && dw2.is_contiguous()
                && dw3.is_contiguous(), "mlp3_wgrad shapes");
    const c10::cuda::CUDAGuard guard(dz1.device());
    float* bptr = nullptr;
    if (bias_scratch.has_value()) {
        CHECK_CONT(*bias_scratch);
        TORCH_CHECK(bias_scratch->numel() >= 3 * H
                    && bias_scratch->dtype() == torch::kFloat32,
                    "bias_scratch layout");
        bptr = bias_scratch->data_ptr<float>();
    }
    emb_mlp3_wgrad(dz1.data_ptr(), dz2.data_ptr(), dz3.data_ptr(),
                   x0.data_ptr(), a1.data_ptr(), a2.data_ptr(), M, H, K0p,
                   K0, scratch.data_ptr<float>(), dw1.data_ptr(),
                   dw2.data_ptr(), dw3.data_ptr(), bptr, cur_stream());
}

void mlp3_pack(torch::Tensor s1, torch::Tensor d1, bool t1,
               torch::Tensor s2, torch::Tensor d2, bool t2,
               torch::Tensor s3, torch::Tensor d3, bool t3) {
    // one launch refreshing three padded weight copies (plain or
    // transposed); src is the [rows, cols] weight, dst the padded buffer
    torch::Tensor ss[3] = {s1, s2, s3}, dd[3] = {d1, d2, d3};
    long a[3][6];
    int tt[3] = {t1, t2, t3};
    for (int i = 0; i < 3; i++) {
        CHECK_GPU(ss[i]); CHECK_CONT(ss[i]);
        TORCH_CHECK(ss[i].dtype() == torch::kBFloat16
                    && dd[i].dtype() == torch::kBFloat16
                    && ss[i].dim() == 2 && dd[i].dim() == 2
                    && dd[i].stride(1) == 1, "mlp3_pack dtypes/shapes");
        long r = ss[i].size(0), c = ss[i].size(1);
        long dr = tt[i] ? c : r, dc = tt[i] ? r : c;
        TORCH_CHECK(dd[i].size(0) >= dr && dd[i].size(1) >= dc,
                    "mlp3_pack: dst too small");
        a[i][0] = r; a[i][1] = c;
        a[i][2] = ss[i].stride(0); a[i][3] = dd[i].stride(0);
    }
    const c10::cuda::CUDAGuard guard(s1.device());
    emb_mlp3_pack(s1.data_ptr(), d1.data_ptr(), a[0][0], a[0][1], a[0][2],
                  a[0][3], tt[0],
                  s2.data_ptr(), d2.data_ptr(), a[1][0], a[1][1], a[1][2],
                  a[1][3], tt[1],
                  s3.data_ptr(), d3.data_ptr(), a[2][0], a[2][1], a[2][2],
                  a[2][3], tt[2], cur_stream());
}

// ---- fused BCE-with-logits --------------------------------------------

torch::Tensor bce_fwd(torch::Tensor logits, torch::Tensor labels) {
    CHECK_GPU(logits); CHECK_CONT(logits); CHECK_CONT(labels);
    TORCH_CHECK(logits.dtype() == torch::kFloat32 &&
                labels.dtype() == torch::kFloat32, "bce_fwd wants fp32");
    TORCH_CHECK(logits.numel() == labels.numel(), "logits/labels mismatch");
    const c10::cuda::CUDAGuard guard(logits.device());
    auto loss = torch::empty({}, logits.options());
    emb_bce_fwd(logits.data_ptr<float>(), labels.data_ptr<float>(),
                logits.numel(), loss.data_ptr<float>(), cur_stream());
    return loss;
}

torch::Tensor bce_bwd(torch::Tensor logits, torch::Tensor labels,
                      torch::Tensor grad_out) {
    CHECK_GPU(logits); CHECK_CONT(logits); CHECK_CONT(labels);
    TORCH_CHECK(grad_out.numel() == 1, "grad_out must be scalar");
    const c10::cuda::CUDAGuard guard(logits.device());
    auto g = torch::empty_like(logits);
    emb_bce_bwd(logits.data_ptr<float>(), labels.data_ptr<float>(),
                logits.numel(), grad_out.contiguous().data_ptr<float>(),
                g.data_ptr<float>(), cur_stream());
    return g;
}

// ---- fused 3-layer MLP forward ----------------------------------------

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
mlp3_fwd(torch::Tensor x0, torch::Tensor w1, torch::Tensor b1,
         torch::Tensor w2, torch::Tensor b2, torch::Tensor w3,
         torch::Tensor b3, torch::Tensor w4, torch::Tensor b4,
         c10::optional<torch::Tensor> partial) {
    CHECK_GPU(x0); CHECK_CONT(x0);
    TORCH_CHECK(x0.dtype() == torch::kBFloat16, "mlp3_fwd: x0 must be bf16");
    const c10::cuda::CUDAGuard guard(x0.device());
    long M = x0.size(0), K0p = x0.size(1), H = w1.size(0);
    long Hp = w2.size(1);
    TORCH_CHECK(H % 16 == 0 && H <= 416, "mlp3_fwd: H must be <=416, x16");
    TORCH_CHECK(K0p % 32 == 0 && Hp % 32 == 0,
                "mlp3_fwd: padded K dims must be x32");
    TORCH_CHECK(w1.size(1) == K0p && w2.size(0) == H && w3.size(0) == H
                && w3.size(1) == Hp && w4.numel() == H);
    auto a1 = torch::empty({M, H}, x0.options());
    auto a2 = torch::empty({M, H}, x0.options());
    auto a3 = torch::empty({M, H}, x0.options());
    auto out = torch::empty({M}, x0.options().dtype(torch::kFloat32));
    const float* pp = nullptr;
    if (partial.has_value()) {
        TORCH_CHECK(partial->is_contiguous() && partial->numel() == M &&
                    partial->dtype() == torch::kFloat32,
                    "mlp3_fwd: partial must be contiguous fp32 [M]");
        pp = partial->data_ptr<float>();
    }
    emb_mlp3_fwd(x0.data_ptr(), M, K0p,
                 w1.data_ptr(), b1.data_ptr(), w2.data_ptr(), b2.data_ptr(),
                 w3.data_ptr(), b3.data_ptr(), w4.data_ptr(), b4.data_ptr(),
                 pp, H, Hp, a1.data_ptr(), a2.data_ptr(), a3.data_ptr(),
                 out.data_ptr<float>(), cur_stream());
    return {out, a1, a2, a3};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
mlp3_bwd(torch::Tensor dout, torch::Tensor a1, torch::Tensor a2,
         torch::Tensor a3, torch::Tensor w4, torch::Tensor w3t,
         torch::Tensor w2t, torch::Tensor w1t, OptTensor bias_scratch) {
    // bias_scratch ([4H+1] fp32): when given, the dz3 assembly loop also
    // accumulates dw4 (dout.a3 column sums) and db4 into segments 3H/4H
    CHECK_GPU(dout); CHECK_CONT(dout); CHECK_CONT(w3t); CHECK_CONT(w2t);
    CHECK_CONT(w1t);
    const c10::cuda::CUDAGuard guard(dout.device());
    long M = a1.size(0), H = a1.size(1), K0p = w1t.size(0);
    long Hp = w3t.size(1);
    TORCH_CHECK(Hp % 32 == 0 && w2t.size(1) == Hp && w1t.size(1) == Hp);
    auto dz1 = torch::empty_like(a1);
    auto dz2 = torch::empty_like(a2);
    auto dz3 = torch::empty_like(a3);
    auto dx0 = torch::empty({M, K0p}, a1.options());
    float* bptr = nullptr;
    if (bias_scratch.has_value()) {
        CHECK_CONT(*bias_scratch);
        TORCH_CHECK(bias_scratch->numel() >= 4 * H + 1
                    && bias_scratch->dtype() == torch::kFloat32,
                    "bias_scratch layout");
        bptr = bias_scratch->data_ptr<float>();
    }
    emb_mlp3_bwd(dout.data_ptr<float>(), M, K0p, a1.data_ptr(),
                 a2.data_ptr(), a3.data_ptr(), w4.data_ptr(), w3t.data_ptr(),
                 w2t.data_ptr(), w1t.data_ptr(), H, Hp, dz1.data_ptr(),
                 dz2.data_ptr(), dz3.data_ptr(), dx0.data_ptr(), bptr,
                 cur_stream());
    return {dx0, dz1, dz2, dz3};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("unique_inverse", &unique_inverse, "hash-based unique+inverse");
    m.def("unique_bounded", &unique_bounded,
          "unique+inverse without host sync (n-sized buffer + device "
          "count); optional fused per-field key offsets",
          pybind11::arg("keys"),
          pybind11::arg("field_offsets") = pybind11::none());
    m.def("ht_lookup", &ht_lookup, "hash table lookup/insert");
    m.def("ht_rehash", &ht_rehash, "hash table rehash into larger table");
    m.def("array_touch", &array_touch,
          "array-table slot compute + valid-bitmap touch");
    m.def("gather_init", &gather_init,
          "row gather with fused lazy init (+ optional duplicate scatter)");
    m.def("reduce_by_inverse", &reduce_by_inverse,
          "grad reduce-by-key with counts (LDS-aggregated)");
    m.def("apply_optimizer", &apply_optimizer, "fused sparse optimizer step");
    m.def("mask_tail", &mask_tail,
          "zero the garbage tail of a bounded block (enables multi-block "
          "merge by concatenation)");
    m.def("fault_in", &fault_in,
          "tier v2: copy host-resident rows into fresh cache slots, "
          "clearing their lazy-init mask");
    m.def("spill_rows", &spill_rows,
          "tier v2: copy cache rows into the pinned host slab");
    m.def("gather_host", &gather_host,
          "tier v2: read-only host-row gather for cache misses");
    m.def("bucketize_pad", &bucketize_pad,
          "owner-bucketize unique keys into a fixed padded [world, cap] "
          "wire block (sync-free multi-rank route)");
    m.def("gather_pad", &gather_pad,
          "gather grads+counts payload into the padded send layout");
    m.def("scatter_out", &scatter_out,
          "padded pull: wire rows -> per-element output (fused dup scatter)");
    m.def("split_payload", &split_payload,
          "split owner-reduced payload into grads + int64 counts");
    m.def("ctr_head_fwd", &ctr_head_fwd,
          "fused CTR head fwd: deep_in assembly (+cast) + FM + first-order "
          "+ dense linear");
    m.def("ctr_head_bwd", &ctr_head_bwd, "fused CTR head backward");
    m.def("mlp3_bwd", &mlp3_bwd,
          "fused dgrad chain backward of the 3-layer MLP (optionally "
          "carrying the head wgrad/bias sums)",
          pybind11::arg("dout"), pybind11::arg("a1"), pybind11::arg("a2"),
          pybind11::arg("a3"), pybind11::arg("w4"), pybind11::arg("w3t"),
          pybind11::arg("w2t"), pybind11::arg("w1t"),
          pybind11::arg("bias_scratch") = pybind11::none());
    m.def("mlp3_pack", &mlp3_pack,
          "refresh three padded (or transposed) weight copies in one "
          "launch");
    m.def("mlp3_fwd", &mlp3_fwd,
          "fused 3-hidden-layer MLP forward (bf16 MFMA, bias+ReLU fused)");
    m.def("mlp3_wgrad", &mlp3_wgrad,
          pybind11::arg("dz1"), pybind11::arg("dz2"), pybind11::arg("dz3"),
          pybind11::arg("x0"), pybind11::arg("a1"), pybind11::arg("a2"),
          pybind11::arg("scratch"), pybind11::arg("dw1"),
          pybind11::arg("dw2"), pybind11::arg("dw3"),
          pybind11::arg("bias_scratch") = pybind11::none(),
          "all three MLP weight grads in one MFMA launch (+= into bf16 "
          "grads via fp32 scratch)");
    m.def("mlp3_bias_bwd", &mlp3_bias_bwd,
          "MLP bias grads + head wgrad in one pass over the dz mirrors "
          "(with_dz=false when the fused wgrad carried the dz sums)",
          pybind11::arg("dout"), pybind11::arg("dz1"), pybind11::arg("dz2"),
          pybind11::arg("dz3"), pybind11::arg("a3"),
          pybind11::arg("scratch"), pybind11::arg("db1"),
          pybind11::arg("db2"), pybind11::arg("db3"), pybind11::arg("dw4"),
          pybind11::arg("db4"), pybind11::arg("with_dz") = true,
          pybind11::arg("with_head") = true);
    m.def("cin_fwd", &cin_fwd,
          "CIN layer forward: implicit outer-product MFMA GEMM");
    m.def("cin_dw", &cin_dw,
          "CIN weight grad: per-field split-K MFMA with on-the-fly operand");
    m.def("cin_dx", &cin_dx,
          "CIN input grads: P=W^T dZ GEMM fused with the xk/x0 contractions");
    m.def("bce_fwd", &bce_fwd, "fused BCE-with-logits forward (mean)");
    m.def("bce_bwd", &bce_bwd, "fused BCE-with-logits backward");
    m.def("flat_adagrad", &flat_adagrad,
          "fused flat-buffer Adagrad (f32, or bf16 weights + f32 master)");
    m.def("flat_opt", &flat_opt,
          "fused flat-buffer dense optimizer (0=sgd, 1=adagrad, 2=adam)");
    m.def("flat_step_scalars", &flat_step_scalars,
          "device-side step counter + Adam bias-correction factors "
          "(capture-safe)");
}
