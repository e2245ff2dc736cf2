/* nezha — public C ABI.
 *
 * This flat nz_* surface is the public interface of the framework: the seam
 * the Go mirror (go/nezha, cgo) binds, and what the Python ctypes test
 * harness (py/nezha_harness.py, nezha_amd/) drives. Opaque handles, int
 * status codes (0 = ok, nonzero = error; nz_last_error() has the message).
 */
#ifndef NEZHA_H_
#define NEZHA_H_

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef int32_t nz_status; /* 0 = ok */
const char* nz_last_error(void);
const char* nz_version(void);

/* dtypes / devices */
#define NZ_F32 0
#define NZ_BF16 1
#define NZ_CPU 0
#define NZ_HIP 1

/* ------------------------------------------------------------- context -- */
typedef struct nz_ctx nz_ctx;
nz_ctx* nz_ctx_new(int device, int device_index);
void nz_ctx_free(nz_ctx*);
nz_status nz_ctx_sync(nz_ctx*);
int nz_hip_device_count(void);
/* allocator high-water mark in bytes */
int64_t nz_ctx_mem_high_water(nz_ctx*);
/* write the NEZHA_TRACE chrome-trace file, if tracing is enabled */
nz_status nz_prof_flush(nz_ctx*);

/* ------------------------------------------------------------- tensors -- */
typedef struct nz_tensor nz_tensor;
nz_tensor* nz_tensor_new(nz_ctx*, int dtype, int ndim, const int64_t* shape);
void nz_tensor_free(nz_tensor*);
int64_t nz_tensor_numel(const nz_tensor*);
int nz_tensor_dtype(const nz_tensor*);
nz_status nz_tensor_upload(nz_tensor*, const void* src, size_t bytes);
nz_status nz_tensor_download(const nz_tensor*, void* dst, size_t bytes);
nz_status nz_tensor_fill(nz_tensor*, float value);
nz_status nz_tensor_rand_uniform(nz_tensor*, uint64_t seed, float lo, float hi);
nz_status nz_tensor_rand_normal(nz_tensor*, uint64_t seed, float mean, float std);

/* ------------------------------------------------- raw ops (test/bench) -- */
nz_status nz_op_gemm(nz_tensor* A, nz_tensor* B, nz_tensor* C, int transA,
                     int transB, float alpha, float beta, nz_tensor* bias,
                     int fuse_gelu, nz_tensor* save_pre);
nz_status nz_op_gemm_batched(nz_tensor* A, nz_tensor* B, nz_tensor* C,
                             int transA, int transB, float alpha, int64_t batch,
                             int64_t M, int64_t N, int64_t K, int64_t sA,
                             int64_t sB, int64_t sC);
nz_status nz_op_bias_gelu_fwd(nz_tensor* x, nz_tensor* bias, nz_tensor* y,
                              nz_tensor* save_pre);
nz_status nz_op_bias_gelu_bwd(nz_tensor* dy, nz_tensor* z, nz_tensor* dz,
                              nz_tensor* dbias);
nz_status nz_op_layernorm_fwd(nz_tensor* x, nz_tensor* gamma, nz_tensor* beta,
                              nz_tensor* y, nz_tensor* mean, nz_tensor* rstd,
                              float eps);
nz_status nz_op_layernorm_bwd(nz_tensor* dy, nz_tensor* x, nz_tensor* gamma,
                              nz_tensor* mean, nz_tensor* rstd, nz_tensor* dx,
                              nz_tensor* dgamma, nz_tensor* dbeta, float beta_dx);
nz_status nz_op_softmax_fwd(nz_tensor* x, nz_tensor* y, float scale);
nz_status nz_op_softmax_bwd(nz_tensor* dy, nz_tensor* y, nz_tensor* dx, float scale);
nz_status nz_op_adam_step(nz_tensor* param, nz_tensor* grad, nz_tensor* master,
                          nz_tensor* m, nz_tensor* v, float lr, float beta1,
                          float beta2, float eps, int step, float grad_scale,
                          float weight_decay);
nz_status nz_op_sgd_step(nz_tensor* param, nz_tensor* grad, nz_tensor* master,
                         nz_tensor* momentum, float lr, float mu, float grad_scale);
nz_status nz_op_dropout_fwd(nz_tensor* x, nz_tensor* y, nz_tensor* mask,
                            float p, uint64_t seed);
nz_status nz_op_dropout_bwd(nz_tensor* dy, nz_tensor* mask, nz_tensor* dx, float p);
nz_status nz_op_mse_loss(nz_tensor* y, nz_tensor* target, nz_tensor* loss,
                         nz_tensor* dy, float grad_scale);
nz_status nz_op_ce_loss(nz_tensor* logits, nz_tensor* labels, nz_tensor* loss,
                        nz_tensor* dlogits, float grad_scale);
nz_status nz_op_add_out(nz_tensor* out, nz_tensor* x, nz_tensor* y);
nz_status nz_op_add_inplace(nz_tensor* dst, nz_tensor* src, float alpha);
nz_status nz_op_scale_inplace(nz_tensor* dst, float alpha);
nz_status nz_op_cast_copy(nz_tensor* dst, nz_tensor* src);
nz_status nz_op_transpose2d(nz_tensor* src, nz_tensor* dst);
nz_status nz_op_colsum(nz_tensor* src, nz_tensor* dst, float beta);
nz_status nz_op_reduce_sum(nz_tensor* src, nz_tensor* out);
nz_status nz_op_l2norm_sq(nz_tensor* src, nz_tensor* out);
nz_status nz_op_attn_split_qkv(nz_tensor* qkv, nz_tensor* q, nz_tensor* k,
                               nz_tensor* v, int64_t B, int64_t S, int64_t heads,
                               int64_t dh);
nz_status nz_op_attn_fused_fwd(nz_tensor* q, nz_tensor* k, nz_tensor* v,
                               nz_tensor* probs, nz_tensor* ctx_out,
                               int64_t batch_heads, int64_t S, int64_t dh,
                               float scale);
nz_status nz_op_attn_fused_bwd(nz_tensor* q, nz_tensor* k, nz_tensor* v,
                               nz_tensor* probs, nz_tensor* dctx, nz_tensor* dq,
                               nz_tensor* dk, nz_tensor* dv,
                               int64_t batch_heads, int64_t S, int64_t dh,
                               float scale);
nz_status nz_op_attn_fwd_packed(nz_tensor* qkv, nz_tensor* probs, nz_tensor* out,
                                int64_t B, int64_t S, int64_t heads, int64_t dh,
                                float scale);
nz_status nz_op_attn_bwd_packed(nz_tensor* qkv, nz_tensor* probs,
                                nz_tensor* dmerged, nz_tensor* dqkv, int64_t B,
                                int64_t S, int64_t heads, int64_t dh,
                                float scale);
nz_status nz_op_attn_merge(nz_tensor* ctx, nz_tensor* out, int64_t B, int64_t S,
                           int64_t heads, int64_t dh);

/* ---------------------------------------------------------------- comm -- */
typedef struct nz_comm nz_comm;
int nz_comm_uid_bytes(void);
nz_status nz_comm_get_uid(void* out);
nz_comm* nz_comm_init_rccl(nz_ctx*, int nranks, int rank, const void* uid);
nz_comm* nz_comm_init_tcp(int nranks, int rank, const char* host, int port);
void nz_comm_free(nz_comm*);
int nz_comm_rank(nz_comm*);
int nz_comm_nranks(nz_comm*);
nz_status nz_comm_barrier(nz_comm*);
/* op: 0 = sum, 1 = max */
nz_status nz_comm_allreduce_scalar(nz_comm*, float* inout, int op);
nz_status nz_comm_allreduce_tensor(nz_comm*, nz_ctx*, nz_tensor*, int op);
nz_status nz_comm_broadcast_tensor(nz_comm*, nz_ctx*, nz_tensor*, int root);

/* ------------------------------------------------------ models/training -- */
typedef struct nz_trainer nz_trainer;
/* optim: "adam" | "sgd". bucket_bytes <= 0 -> default 64 MiB. */
nz_trainer* nz_trainer_new_mlp(nz_ctx*, int64_t layers, int64_t dim,
                               int64_t batch, const char* optim, float lr,
                               int64_t bucket_bytes, uint64_t seed);
nz_trainer* nz_trainer_new_bert(nz_ctx*, int64_t layers, int64_t hidden,
                                int64_t ffn, int64_t heads, int64_t seq,
                                int64_t batch, const char* optim, float lr,
                                int64_t bucket_bytes, uint64_t seed);
void nz_trainer_free(nz_trainer*);
int64_t nz_trainer_num_params(nz_trainer*);
nz_status nz_trainer_attach_ddp(nz_trainer*, nz_comm*);
nz_status nz_trainer_reseed_data(nz_trainer*, uint64_t seed);
nz_status nz_trainer_step(nz_trainer*);           /* async on HIP */
nz_status nz_trainer_loss(nz_trainer*, float* out); /* syncs */
int nz_trainer_step_count(nz_trainer*);
int64_t nz_trainer_samples_per_step(nz_trainer*);
double nz_trainer_flops_per_step(nz_trainer*);

/* --------------------------------------------------------- checkpointing -- */
nz_status nz_ckpt_save(nz_trainer*, const char* path);
nz_status nz_ckpt_load(nz_trainer*, const char* path);
/* this rank's state to its own file; no rank gating, no collective */
nz_status nz_ckpt_save_local(nz_trainer*, const char* path);

#ifdef __cplusplus
} /* extern "C" */
#endif
#endif /* NEZHA_H_ */
