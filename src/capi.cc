// C ABI implementation: thin exception-catching
// wrappers from nz_* entry points onto the C++ engine.
#include "nezha.h"

#include <memory>

#include "src/comm/comm.h"
#include "src/models/models.h"
#include "src/train/trainer.h"
#include "src/util/prof.h"
#include "src/util/util.h"

using namespace nz;

static thread_local std::string g_err;

extern "C" const char* nz_last_error(void) { return g_err.c_str(); }
extern "C" const char* nz_version(void) { return "nezha-amd 0.1.0 (gfx950)"; }

#define NZ_API_BEGIN try {
#define NZ_API_END                       \
  return 0;                              \
  }                                      \
  catch (const std::exception& e) {      \
    g_err = e.what();                    \
    return 1;                            \
  }
#define NZ_API_END_PTR(T)                \
  }                                      \
  catch (const std::exception& e) {      \
    g_err = e.what();                    \
    return (T) nullptr;                  \
  }

struct nz_ctx {
  Context impl;
  nz_ctx(Device d, int i) : impl(d, i) {}
};
struct nz_tensor {
  Tensor t;
};
struct nz_comm {
  std::unique_ptr<Comm> c;
};
struct nz_trainer {
  std::unique_ptr<Trainer> tr;
};

// ------------------------------------------------------------------ ctx ---
extern "C" nz_ctx* nz_ctx_new(int device, int device_index) {
  NZ_API_BEGIN
  return new nz_ctx(device == NZ_HIP ? Device::HIP : Device::CPU, device_index);
  NZ_API_END_PTR(nz_ctx*)
}
extern "C" void nz_ctx_free(nz_ctx* c) { delete c; }
extern "C" nz_status nz_ctx_sync(nz_ctx* c) {
  NZ_API_BEGIN
  c->impl.sync();
  NZ_API_END
}
extern "C" int nz_hip_device_count(void) { return hip_device_count(); }
extern "C" nz_status nz_prof_flush(nz_ctx* c) {
  NZ_API_BEGIN
  nz::prof::flush(&c->impl);
  NZ_API_END
}
extern "C" int64_t nz_ctx_mem_high_water(nz_ctx* c) {
  return (int64_t)c->impl.arena()->high_water();
}

// --------------------------------------------------------------- tensors ---
extern "C" nz_tensor* nz_tensor_new(nz_ctx* c, int dtype, int ndim,
                                    const int64_t* shape) {
  NZ_API_BEGIN
  Shape s;
  for (int i = 0; i < ndim; ++i) s.dims.push_back(shape[i]);
  return new nz_tensor{
      Tensor(&c->impl, dtype == NZ_BF16 ? DType::BF16 : DType::F32, s)};
  NZ_API_END_PTR(nz_tensor*)
}
extern "C" void nz_tensor_free(nz_tensor* t) { delete t; }
extern "C" int64_t nz_tensor_numel(const nz_tensor* t) { return t->t.numel(); }
extern "C" int nz_tensor_dtype(const nz_tensor* t) {
  return t->t.dtype() == DType::BF16 ? NZ_BF16 : NZ_F32;
}
extern "C" nz_status nz_tensor_upload(nz_tensor* t, const void* src, size_t n) {
  NZ_API_BEGIN
  t->t.upload(src, n);
  NZ_API_END
}
extern "C" nz_status nz_tensor_download(const nz_tensor* t, void* dst, size_t n) {
  NZ_API_BEGIN
  t->t.download(dst, n);
  NZ_API_END
}
extern "C" nz_status nz_tensor_fill(nz_tensor* t, float v) {
  NZ_API_BEGIN
  ops::fill(t->t, v);
  NZ_API_END
}
extern "C" nz_status nz_tensor_rand_uniform(nz_tensor* t, uint64_t seed, float lo,
                                            float hi) {
  NZ_API_BEGIN
  ops::rand_uniform(t->t, seed, lo, hi);
  NZ_API_END
}
extern "C" nz_status nz_tensor_rand_normal(nz_tensor* t, uint64_t seed, float mean,
                                           float std) {
  NZ_API_BEGIN
  ops::rand_normal(t->t, seed, mean, std);
  NZ_API_END
}

// ------------------------------------------------------------------- ops ---
extern "C" nz_status nz_op_gemm(nz_tensor* A, nz_tensor* B, nz_tensor* C,
                                int transA, int transB, float alpha, float beta,
                                nz_tensor* bias, int fuse_gelu,
                                nz_tensor* save_pre) {
  NZ_API_BEGIN
  ops::GemmEpilogue epi;
  epi.bias = bias ? &bias->t : nullptr;
  epi.gelu = fuse_gelu != 0;
  epi.save_pre = save_pre ? &save_pre->t : nullptr;
  bool has_epi = epi.bias || epi.gelu || epi.save_pre;
  ops::gemm(A->t, B->t, C->t, transA != 0, transB != 0, alpha, beta,
            has_epi ? &epi : nullptr);
  NZ_API_END
}
extern "C" nz_status nz_op_gemm_batched(nz_tensor* A, nz_tensor* B, nz_tensor* C,
                                        int transA, int transB, float alpha,
                                        int64_t batch, int64_t M, int64_t N,
                                        int64_t K, int64_t sA, int64_t sB,
                                        int64_t sC) {
  NZ_API_BEGIN
  ops::gemm_batched(A->t, B->t, C->t, transA != 0, transB != 0, alpha, batch, M,
                    N, K, sA, sB, sC);
  NZ_API_END
}
extern "C" nz_status nz_op_bias_gelu_fwd(nz_tensor* x, nz_tensor* bias,
                                         nz_tensor* y, nz_tensor* save_pre) {
  NZ_API_BEGIN
  ops::bias_gelu_fwd(x->t, bias ? &bias->t : nullptr, y->t,
                     save_pre ? &save_pre->t : nullptr);
  NZ_API_END
}
extern "C" nz_status nz_op_bias_gelu_bwd(nz_tensor* dy, nz_tensor* z,
                                         nz_tensor* dz, nz_tensor* dbias) {
  NZ_API_BEGIN
  ops::bias_gelu_bwd(dy->t, z->t, dz->t, dbias ? &dbias->t : nullptr);
  NZ_API_END
}
extern "C" nz_status nz_op_layernorm_fwd(nz_tensor* x, nz_tensor* gamma,
                                         nz_tensor* beta, nz_tensor* y,
                                         nz_tensor* mean, nz_tensor* rstd,
                                         float eps) {
  NZ_API_BEGIN
  ops::layernorm_fwd(x->t, gamma->t, beta->t, y->t, mean->t, rstd->t, eps);
  NZ_API_END
}
extern "C" nz_status nz_op_layernorm_bwd(nz_tensor* dy, nz_tensor* x,
                                         nz_tensor* gamma, nz_tensor* mean,
                                         nz_tensor* rstd, nz_tensor* dx,
                                         nz_tensor* dgamma, nz_tensor* dbeta,
                                         float beta_dx) {
  NZ_API_BEGIN
  ops::layernorm_bwd(dy->t, x->t, gamma->t, mean->t, rstd->t, dx->t, dgamma->t,
                     dbeta->t, beta_dx);
  NZ_API_END
}
extern "C" nz_status nz_op_softmax_fwd(nz_tensor* x, nz_tensor* y, float scale) {
  NZ_API_BEGIN
  ops::softmax_fwd(x->t, y->t, scale);
  NZ_API_END
}
extern "C" nz_status nz_op_softmax_bwd(nz_tensor* dy, nz_tensor* y, nz_tensor* dx,
                                       float scale) {
  NZ_API_BEGIN
  ops::softmax_bwd(dy->t, y->t, dx->t, scale);
  NZ_API_END
}
extern "C" nz_status nz_op_adam_step(nz_tensor* param, nz_tensor* grad,
                                     nz_tensor* master, nz_tensor* m, nz_tensor* v,
                                     float lr, float b1, float b2, float eps,
                                     int step, float gs, float wd) {
  NZ_API_BEGIN
  ops::adam_step(param->t, grad->t, master->t, m->t, v->t, lr, b1, b2, eps, step,
                 gs, wd);
  NZ_API_END
}
extern "C" nz_status nz_op_sgd_step(nz_tensor* param, nz_tensor* grad,
                                    nz_tensor* master, nz_tensor* momentum,
                                    float lr, float mu, float gs) {
  NZ_API_BEGIN
  ops::sgd_step(param->t, grad->t, master->t, momentum ? &momentum->t : nullptr,
                lr, mu, gs);
  NZ_API_END
}
extern "C" nz_status nz_op_dropout_fwd(nz_tensor* x, nz_tensor* y,
                                       nz_tensor* mask, float p, uint64_t seed) {
  NZ_API_BEGIN
  ops::dropout_fwd(x->t, y->t, mask->t, p, seed);
  NZ_API_END
}
extern "C" nz_status nz_op_dropout_bwd(nz_tensor* dy, nz_tensor* mask,
                                       nz_tensor* dx, float p) {
  NZ_API_BEGIN
  ops::dropout_bwd(dy->t, mask->t, dx->t, p);
  NZ_API_END
}
extern "C" nz_status nz_op_mse_loss(nz_tensor* y, nz_tensor* target,
                                    nz_tensor* loss, nz_tensor* dy, float gs) {
  NZ_API_BEGIN
  ops::mse_loss(y->t, target->t, loss->t, dy ? &dy->t : nullptr, gs);
  NZ_API_END
}
extern "C" nz_status nz_op_ce_loss(nz_tensor* logits, nz_tensor* labels,
                                   nz_tensor* loss, nz_tensor* dlogits, float gs) {
  NZ_API_BEGIN
  ops::ce_loss(logits->t, labels->t, loss->t, dlogits ? &dlogits->t : nullptr, gs);
  NZ_API_END
}
extern "C" nz_status nz_op_add_out(nz_tensor* out, nz_tensor* x, nz_tensor* y) {
  NZ_API_BEGIN
  ops::add_out(out->t, x->t, y->t);
  NZ_API_END
}
extern "C" nz_status nz_op_add_inplace(nz_tensor* dst, nz_tensor* src, float a) {
  NZ_API_BEGIN
  ops::add_inplace(dst->t, src->t, a);
  NZ_API_END
}
extern "C" nz_status nz_op_scale_inplace(nz_tensor* dst, float a) {
  NZ_API_BEGIN
  ops::scale_inplace(dst->t, a);
  NZ_API_END
}
extern "C" nz_status nz_op_cast_copy(nz_tensor* dst, nz_tensor* src) {
  NZ_API_BEGIN
  ops::cast_copy(dst->t, src->t);
  NZ_API_END
}
extern "C" nz_status nz_op_transpose2d(nz_tensor* src, nz_tensor* dst) {
  NZ_API_BEGIN
  ops::transpose2d(src->t, dst->t);
  NZ_API_END
}
extern "C" nz_status nz_op_colsum(nz_tensor* src, nz_tensor* dst, float beta) {
  NZ_API_BEGIN
  ops::colsum(src->t, dst->t, beta);
  NZ_API_END
}
extern "C" nz_status nz_op_reduce_sum(nz_tensor* src, nz_tensor* out) {
  NZ_API_BEGIN
  ops::reduce_sum(src->t, out->t);
  NZ_API_END
}
extern "C" nz_status nz_op_l2norm_sq(nz_tensor* src, nz_tensor* out) {
  NZ_API_BEGIN
  ops::l2norm_sq(src->t, out->t);
  NZ_API_END
}
extern "C" nz_status nz_op_attn_split_qkv(nz_tensor* qkv, nz_tensor* q,
                                          nz_tensor* k, nz_tensor* v, int64_t B,
                                          int64_t S, int64_t heads, int64_t dh) {
  NZ_API_BEGIN
  ops::attn_split_qkv(qkv->t, q->t, k->t, v->t, B, S, heads, dh);
  NZ_API_END
}
extern "C" nz_status nz_op_attn_fused_fwd(nz_tensor* q, nz_tensor* k,
                                          nz_tensor* v, nz_tensor* probs,
                                          nz_tensor* ctx_out,
                                          int64_t batch_heads, int64_t S,
                                          int64_t dh, float scale) {
  NZ_API_BEGIN
  ops::attn_fused_fwd(q->t, k->t, v->t, probs->t, ctx_out->t, batch_heads, S,
                      dh, scale);
  NZ_API_END
}
extern "C" nz_status nz_op_attn_fused_bwd(nz_tensor* q, nz_tensor* k,
                                          nz_tensor* v, nz_tensor* probs,
                                          nz_tensor* dctx, nz_tensor* dq,
                                          nz_tensor* dk, nz_tensor* dv,
                                          int64_t batch_heads, int64_t S,
                                          int64_t dh, float scale) {
  NZ_API_BEGIN
  ops::attn_fused_bwd(q->t, k->t, v->t, probs->t, dctx->t, dq->t, dk->t,
                      dv->t, batch_heads, S, dh, scale);
  NZ_API_END
}
extern "C" nz_status nz_op_attn_fwd_packed(nz_tensor* qkv, nz_tensor* probs,
                                           nz_tensor* out, int64_t B, int64_t S,
                                           int64_t heads, int64_t dh,
                                           float scale) {
  NZ_API_BEGIN
  ops::attn_fwd_packed(qkv->t, probs->t, out->t, B, S, heads, dh, scale);
  NZ_API_END
}
extern "C" nz_status nz_op_attn_bwd_packed(nz_tensor* qkv, nz_tensor* probs,
                                           nz_tensor* dmerged, nz_tensor* dqkv,
                                           int64_t B, int64_t S, int64_t heads,
                                           int64_t dh, float scale) {
  NZ_API_BEGIN
  ops::attn_bwd_packed(qkv->t, probs->t, dmerged->t, dqkv->t, B, S, heads, dh,
                       scale);
  NZ_API_END
}
extern "C" nz_status nz_op_attn_merge(nz_tensor* ctx, nz_tensor* out, int64_t B,
                                      int64_t S, int64_t heads, int64_t dh) {
  NZ_API_BEGIN
  ops::attn_merge(ctx->t, out->t, B, S, heads, dh);
  NZ_API_END
}

// ------------------------------------------------------------------ comm ---
extern "C" int nz_comm_uid_bytes(void) { return comm_uid_bytes(); }
extern "C" nz_status nz_comm_get_uid(void* out) {
  NZ_API_BEGIN
  comm_get_uid(out);
  NZ_API_END
}
extern "C" nz_comm* nz_comm_init_rccl(nz_ctx* c, int nranks, int rank,
                                      const void* uid) {
  NZ_API_BEGIN
  return new nz_comm{make_rccl_comm(&c->impl, nranks, rank, uid)};
  NZ_API_END_PTR(nz_comm*)
}
extern "C" nz_comm* nz_comm_init_tcp(int nranks, int rank, const char* host,
                                     int port) {
  NZ_API_BEGIN
  return new nz_comm{make_tcp_comm(nranks, rank, host, port)};
  NZ_API_END_PTR(nz_comm*)
}
extern "C" void nz_comm_free(nz_comm* c) { delete c; }
extern "C" int nz_comm_rank(nz_comm* c) { return c->c->rank(); }
extern "C" int nz_comm_nranks(nz_comm* c) { return c->c->nranks(); }
extern "C" nz_status nz_comm_barrier(nz_comm* c) {
  NZ_API_BEGIN
  c->c->barrier();
  NZ_API_END
}
extern "C" nz_status nz_comm_allreduce_scalar(nz_comm* c, float* inout, int op) {
  NZ_API_BEGIN
  *inout = c->c->all_reduce_scalar(*inout,
                                   op == 1 ? ReduceOp::MAX : ReduceOp::SUM);
  NZ_API_END
}
extern "C" nz_status nz_comm_allreduce_tensor(nz_comm* c, nz_ctx* ctx,
                                              nz_tensor* t, int op) {
  NZ_API_BEGIN
  Stream* s = ctx->impl.device() == Device::HIP ? ctx->impl.compute_stream()
                                                : nullptr;
  c->c->all_reduce(t->t.data(), t->t.numel(), t->t.dtype(),
                   op == 1 ? ReduceOp::MAX : ReduceOp::SUM, s);
  NZ_API_END
}
extern "C" nz_status nz_comm_broadcast_tensor(nz_comm* c, nz_ctx* ctx,
                                              nz_tensor* t, int root) {
  NZ_API_BEGIN
  Stream* s = ctx->impl.device() == Device::HIP ? ctx->impl.compute_stream()
                                                : nullptr;
  c->c->broadcast(t->t.data(), t->t.numel(), t->t.dtype(), root, s);
  NZ_API_END
}

// --------------------------------------------------------------- trainer ---
static Optimizer::Kind parse_optim(const char* s) {
  std::string o = s ? s : "adam";
  NZ_CHECK(o == "adam" || o == "sgd", "unknown optimizer: " + o);
  return o == "adam" ? Optimizer::Kind::ADAM : Optimizer::Kind::SGD;
}

extern "C" nz_trainer* nz_trainer_new_mlp(nz_ctx* c, int64_t layers, int64_t dim,
                                          int64_t batch, const char* optim,
                                          float lr, int64_t bucket_bytes,
                                          uint64_t seed) {
  NZ_API_BEGIN
  auto model = std::make_unique<MlpModel>(&c->impl, layers, dim, batch, seed);
  auto tr = std::make_unique<Trainer>(
      &c->impl, std::move(model), parse_optim(optim), lr,
      bucket_bytes > 0 ? bucket_bytes : (int64_t)64 << 20, seed + 12345);
  return new nz_trainer{std::move(tr)};
  NZ_API_END_PTR(nz_trainer*)
}
extern "C" nz_trainer* nz_trainer_new_bert(nz_ctx* c, int64_t layers,
                                           int64_t hidden, int64_t ffn,
                                           int64_t heads, int64_t seq,
                                           int64_t batch, const char* optim,
                                           float lr, int64_t bucket_bytes,
                                           uint64_t seed) {
  NZ_API_BEGIN
  BertConfig cfg;
  cfg.layers = layers;
  cfg.hidden = hidden;
  cfg.ffn = ffn;
  cfg.heads = heads;
  cfg.seq = seq;
  cfg.batch = batch;
  auto model = std::make_unique<BertEncoderModel>(&c->impl, cfg, seed);
  auto tr = std::make_unique<Trainer>(
      &c->impl, std::move(model), parse_optim(optim), lr,
      bucket_bytes > 0 ? bucket_bytes : (int64_t)64 << 20, seed + 12345);
  return new nz_trainer{std::move(tr)};
  NZ_API_END_PTR(nz_trainer*)
}
extern "C" void nz_trainer_free(nz_trainer* t) { delete t; }
extern "C" int64_t nz_trainer_num_params(nz_trainer* t) {
  return t->tr->model()->num_params();
}
extern "C" nz_status nz_trainer_attach_ddp(nz_trainer* t, nz_comm* c) {
  NZ_API_BEGIN
  t->tr->attach_ddp(c->c.get());
  NZ_API_END
}
extern "C" nz_status nz_trainer_reseed_data(nz_trainer* t, uint64_t seed) {
  NZ_API_BEGIN
  t->tr->reseed_data(seed);
  NZ_API_END
}
extern "C" nz_status nz_trainer_step(nz_trainer* t) {
  NZ_API_BEGIN
  t->tr->step();
  NZ_API_END
}
extern "C" nz_status nz_trainer_loss(nz_trainer* t, float* out) {
  NZ_API_BEGIN
  *out = t->tr->last_loss();
  NZ_API_END
}
extern "C" int nz_trainer_step_count(nz_trainer* t) {
  return t->tr->step_count();
}
extern "C" int64_t nz_trainer_samples_per_step(nz_trainer* t) {
  return t->tr->model()->samples_per_batch();
}
extern "C" double nz_trainer_flops_per_step(nz_trainer* t) {
  return t->tr->model()->flops_per_step();
}

// ------------------------------------------------------------ checkpoint ---
extern "C" nz_status nz_ckpt_save(nz_trainer* t, const char* path) {
  NZ_API_BEGIN
  t->tr->save_checkpoint(path);
  NZ_API_END
}
extern "C" nz_status nz_ckpt_load(nz_trainer* t, const char* path) {
  NZ_API_BEGIN
  t->tr->load_checkpoint(path);
  NZ_API_END
}
/* Per-rank local save (no rank gating / collective): test hook for bitwise
   cross-rank agreement checks under DDP. */
extern "C" nz_status nz_ckpt_save_local(nz_trainer* t, const char* path) {
  NZ_API_BEGIN
  t->tr->save_checkpoint(path, /*local_only=*/true);
  NZ_API_END
}
