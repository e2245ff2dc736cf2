"""nezha_amd — Python binding of the nezha MI355X-native runtime.

This package is a ctypes binding over the public C ABI (include/nezha.h,
implemented by build/libnezha.so). It exists for the pytest harness and the
bench driver: the product path is native C++/HIP —
there is no Python compute here, every op below calls into the hand-written
CDNA4 kernel set (or the CPU f32 reference backend).
"""

import ctypes
import os

import numpy as np

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_LIB_PATHS = [
    os.path.join(_REPO, "build", "libnezha.so"),
    os.path.join(os.path.dirname(os.path.abspath(__file__)), "libnezha.so"),
]

F32, BF16 = 0, 1
CPU, HIP = 0, 1


def _load():
    last = None
    for p in _LIB_PATHS:
        if os.path.exists(p):
            try:
                return ctypes.CDLL(p)
            except OSError as e:  # pragma: no cover
                last = e
    raise RuntimeError(
        f"libnezha.so not found/loadable (tried {_LIB_PATHS}): {last}. "
        "Run `make lib` first."
    )


_lib = _load()

_lib.nz_last_error.restype = ctypes.c_char_p
_lib.nz_version.restype = ctypes.c_char_p
_lib.nz_ctx_new.restype = ctypes.c_void_p
_lib.nz_ctx_new.argtypes = [ctypes.c_int, ctypes.c_int]
_lib.nz_ctx_free.argtypes = [ctypes.c_void_p]
_lib.nz_ctx_sync.argtypes = [ctypes.c_void_p]
_lib.nz_prof_flush.argtypes = [ctypes.c_void_p]
_lib.nz_ctx_mem_high_water.restype = ctypes.c_int64
_lib.nz_ctx_mem_high_water.argtypes = [ctypes.c_void_p]
_lib.nz_tensor_new.restype = ctypes.c_void_p
_lib.nz_tensor_new.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
                               ctypes.POINTER(ctypes.c_int64)]
_lib.nz_tensor_free.argtypes = [ctypes.c_void_p]
_lib.nz_tensor_numel.restype = ctypes.c_int64
_lib.nz_tensor_numel.argtypes = [ctypes.c_void_p]
_lib.nz_tensor_upload.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
_lib.nz_tensor_download.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
_lib.nz_tensor_fill.argtypes = [ctypes.c_void_p, ctypes.c_float]
_lib.nz_tensor_rand_uniform.argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                        ctypes.c_float, ctypes.c_float]
_lib.nz_tensor_rand_normal.argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                       ctypes.c_float, ctypes.c_float]
_lib.nz_op_gemm.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_int, ctypes.c_int,
                            ctypes.c_float, ctypes.c_float, ctypes.c_void_p,
                            ctypes.c_int, ctypes.c_void_p]
_lib.nz_op_gemm_batched.argtypes = [ctypes.c_void_p] * 3 + [
    ctypes.c_int, ctypes.c_int, ctypes.c_float] + [ctypes.c_int64] * 7
_lib.nz_op_bias_gelu_fwd.argtypes = [ctypes.c_void_p] * 4
_lib.nz_op_bias_gelu_bwd.argtypes = [ctypes.c_void_p] * 4
_lib.nz_op_layernorm_fwd.argtypes = [ctypes.c_void_p] * 6 + [ctypes.c_float]
_lib.nz_op_layernorm_bwd.argtypes = [ctypes.c_void_p] * 8 + [ctypes.c_float]
_lib.nz_op_softmax_fwd.argtypes = [ctypes.c_void_p] * 2 + [ctypes.c_float]
_lib.nz_op_softmax_bwd.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_float]
_lib.nz_op_adam_step.argtypes = [ctypes.c_void_p] * 5 + [ctypes.c_float] * 4 + [
    ctypes.c_int, ctypes.c_float, ctypes.c_float]
_lib.nz_op_sgd_step.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_float] * 3
_lib.nz_op_dropout_fwd.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_float, ctypes.c_uint64]
_lib.nz_op_dropout_bwd.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_float]
_lib.nz_op_mse_loss.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_float]
_lib.nz_op_ce_loss.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_float]
_lib.nz_op_add_inplace.argtypes = [ctypes.c_void_p] * 2 + [ctypes.c_float]
_lib.nz_op_add_out.argtypes = [ctypes.c_void_p] * 3
_lib.nz_op_scale_inplace.argtypes = [ctypes.c_void_p, ctypes.c_float]
_lib.nz_op_cast_copy.argtypes = [ctypes.c_void_p] * 2
_lib.nz_op_transpose2d.argtypes = [ctypes.c_void_p] * 2
_lib.nz_op_colsum.argtypes = [ctypes.c_void_p] * 2 + [ctypes.c_float]
_lib.nz_op_reduce_sum.argtypes = [ctypes.c_void_p] * 2
_lib.nz_op_l2norm_sq.argtypes = [ctypes.c_void_p] * 2
_lib.nz_op_attn_split_qkv.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_int64] * 4
_lib.nz_op_attn_fused_fwd.argtypes = [ctypes.c_void_p] * 5 + [ctypes.c_int64] * 3 + [ctypes.c_float]
_lib.nz_op_attn_fused_bwd.argtypes = [ctypes.c_void_p] * 8 + [ctypes.c_int64] * 3 + [ctypes.c_float]
_lib.nz_op_attn_fwd_packed.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_int64] * 4 + [ctypes.c_float]
_lib.nz_op_attn_bwd_packed.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_int64] * 4 + [ctypes.c_float]
_lib.nz_op_attn_merge.argtypes = [ctypes.c_void_p] * 2 + [ctypes.c_int64] * 4
_lib.nz_comm_uid_bytes.restype = ctypes.c_int
_lib.nz_comm_get_uid.argtypes = [ctypes.c_void_p]
_lib.nz_comm_init_rccl.restype = ctypes.c_void_p
_lib.nz_comm_init_rccl.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
                                   ctypes.c_void_p]
_lib.nz_comm_init_tcp.restype = ctypes.c_void_p
_lib.nz_comm_init_tcp.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_char_p,
                                  ctypes.c_int]
_lib.nz_comm_free.argtypes = [ctypes.c_void_p]
_lib.nz_comm_rank.argtypes = [ctypes.c_void_p]
_lib.nz_comm_nranks.argtypes = [ctypes.c_void_p]
_lib.nz_comm_barrier.argtypes = [ctypes.c_void_p]
_lib.nz_comm_allreduce_scalar.argtypes = [ctypes.c_void_p,
                                          ctypes.POINTER(ctypes.c_float),
                                          ctypes.c_int]
_lib.nz_comm_allreduce_tensor.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_int]
_lib.nz_comm_broadcast_tensor.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_int]
_lib.nz_trainer_new_mlp.restype = ctypes.c_void_p
_lib.nz_trainer_new_mlp.argtypes = [ctypes.c_void_p] + [ctypes.c_int64] * 3 + [
    ctypes.c_char_p, ctypes.c_float, ctypes.c_int64, ctypes.c_uint64]
_lib.nz_trainer_new_bert.restype = ctypes.c_void_p
_lib.nz_trainer_new_bert.argtypes = [ctypes.c_void_p] + [ctypes.c_int64] * 6 + [
    ctypes.c_char_p, ctypes.c_float, ctypes.c_int64, ctypes.c_uint64]
_lib.nz_trainer_free.argtypes = [ctypes.c_void_p]
_lib.nz_trainer_num_params.restype = ctypes.c_int64
_lib.nz_trainer_num_params.argtypes = [ctypes.c_void_p]
_lib.nz_trainer_attach_ddp.argtypes = [ctypes.c_void_p] * 2
_lib.nz_trainer_reseed_data.argtypes = [ctypes.c_void_p, ctypes.c_uint64]
_lib.nz_trainer_step.argtypes = [ctypes.c_void_p]
_lib.nz_trainer_loss.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_float)]
_lib.nz_trainer_step_count.argtypes = [ctypes.c_void_p]
_lib.nz_trainer_samples_per_step.restype = ctypes.c_int64
_lib.nz_trainer_samples_per_step.argtypes = [ctypes.c_void_p]
_lib.nz_trainer_flops_per_step.restype = ctypes.c_double
_lib.nz_trainer_flops_per_step.argtypes = [ctypes.c_void_p]
_lib.nz_ckpt_save.argtypes = [ctypes.c_void_p, ctypes.c_char_p]
_lib.nz_ckpt_save_local.argtypes = [ctypes.c_void_p, ctypes.c_char_p]
_lib.nz_ckpt_load.argtypes = [ctypes.c_void_p, ctypes.c_char_p]
_lib.nz_hip_device_count.restype = ctypes.c_int


class NezhaError(RuntimeError):
    pass


def _ck(status):
    if status != 0:
        raise NezhaError(_lib.nz_last_error().decode())


def version():
    return _lib.nz_version().decode()


def hip_device_count():
    return _lib.nz_hip_device_count()


def bf16_to_f32(u16):
    """numpy uint16 bf16 bits -> float32"""
    return (u16.astype(np.uint32) << 16).view(np.float32)


def f32_to_bf16(f32):
    """numpy float32 -> uint16 bf16 bits (round-to-nearest-even)"""
    u = np.asarray(f32, dtype=np.float32).view(np.uint32)
    rounding = np.uint32(0x7FFF) + ((u >> 16) & 1)
    return ((u + rounding) >> 16).astype(np.uint16)


class Ctx:
    def __init__(self, device=CPU, index=0):
        self._h = _lib.nz_ctx_new(device, index)
        if not self._h:
            raise NezhaError(_lib.nz_last_error().decode())
        self.device = device

    def sync(self):
        _ck(_lib.nz_ctx_sync(self._h))

    def mem_high_water(self):
        return _lib.nz_ctx_mem_high_water(self._h)

    def prof_flush(self):
        """write the NEZHA_TRACE chrome-trace file (no-op when disabled)"""
        _ck(_lib.nz_prof_flush(self._h))

    def __del__(self):
        if getattr(self, "_h", None) and _lib:
            _lib.nz_ctx_free(self._h)
            self._h = None

    # -- tensor constructors ------------------------------------------------
    def tensor(self, shape, dtype=F32):
        return Tensor(self, shape, dtype)

    def from_numpy(self, arr):
        """f32 array -> f32 tensor; uint16 array -> bf16 tensor."""
        arr = np.ascontiguousarray(arr)
        if arr.dtype == np.float32:
            t = Tensor(self, arr.shape, F32)
        elif arr.dtype == np.uint16:
            t = Tensor(self, arr.shape, BF16)
        else:
            raise TypeError(f"unsupported dtype {arr.dtype}")
        t.upload(arr)
        return t

    def from_numpy_bf16(self, arr_f32):
        return self.from_numpy(f32_to_bf16(arr_f32))


class Tensor:
    def __init__(self, ctx, shape, dtype):
        self.ctx = ctx
        self.shape = tuple(int(s) for s in shape)
        self.dtype = dtype
        arr = (ctypes.c_int64 * len(self.shape))(*self.shape)
        self._h = _lib.nz_tensor_new(ctx._h, dtype, len(self.shape), arr)
        if not self._h:
            raise NezhaError(_lib.nz_last_error().decode())

    def __del__(self):
        if getattr(self, "_h", None) and _lib:
            _lib.nz_tensor_free(self._h)
            self._h = None

    @property
    def numel(self):
        n = 1
        for s in self.shape:
            n *= s
        return n

    def upload(self, arr):
        arr = np.ascontiguousarray(arr)
        _ck(_lib.nz_tensor_upload(self._h, arr.ctypes.data_as(ctypes.c_void_p),
                                  arr.nbytes))

    def numpy(self):
        """Download. f32 -> float32 array; bf16 -> float32 array (converted)."""
        if self.dtype == F32:
            out = np.empty(self.shape, dtype=np.float32)
            _ck(_lib.nz_tensor_download(self._h,
                                        out.ctypes.data_as(ctypes.c_void_p),
                                        out.nbytes))
            return out
        out = np.empty(self.shape, dtype=np.uint16)
        _ck(_lib.nz_tensor_download(self._h, out.ctypes.data_as(ctypes.c_void_p),
                                    out.nbytes))
        return bf16_to_f32(out)

    def raw(self):
        dt = np.float32 if self.dtype == F32 else np.uint16
        out = np.empty(self.shape, dtype=dt)
        _ck(_lib.nz_tensor_download(self._h, out.ctypes.data_as(ctypes.c_void_p),
                                    out.nbytes))
        return out

    def fill(self, v):
        _ck(_lib.nz_tensor_fill(self._h, v))
        return self

    def rand_uniform(self, seed, lo=0.0, hi=1.0):
        _ck(_lib.nz_tensor_rand_uniform(self._h, seed, lo, hi))
        return self

    def rand_normal(self, seed, mean=0.0, std=1.0):
        _ck(_lib.nz_tensor_rand_normal(self._h, seed, mean, std))
        return self


def _h(t):
    return t._h if t is not None else None


# ------------------------------------------------------------------- ops ---
def gemm(A, B, C, transA=False, transB=False, alpha=1.0, beta=0.0, bias=None,
         gelu=False, save_pre=None):
    _ck(_lib.nz_op_gemm(_h(A), _h(B), _h(C), int(transA), int(transB), alpha,
                        beta, _h(bias), int(gelu), _h(save_pre)))


def gemm_batched(A, B, C, transA, transB, alpha, batch, M, N, K, sA, sB, sC):
    _ck(_lib.nz_op_gemm_batched(_h(A), _h(B), _h(C), int(transA), int(transB),
                                alpha, batch, M, N, K, sA, sB, sC))


def bias_gelu_fwd(x, bias, y, save_pre=None):
    _ck(_lib.nz_op_bias_gelu_fwd(_h(x), _h(bias), _h(y), _h(save_pre)))


def bias_gelu_bwd(dy, z, dz, dbias=None):
    _ck(_lib.nz_op_bias_gelu_bwd(_h(dy), _h(z), _h(dz), _h(dbias)))


def layernorm_fwd(x, gamma, beta, y, mean, rstd, eps=1e-5):
    _ck(_lib.nz_op_layernorm_fwd(_h(x), _h(gamma), _h(beta), _h(y), _h(mean),
                                 _h(rstd), eps))


def layernorm_bwd(dy, x, gamma, mean, rstd, dx, dgamma, dbeta, beta_dx=0.0):
    _ck(_lib.nz_op_layernorm_bwd(_h(dy), _h(x), _h(gamma), _h(mean), _h(rstd),
                                 _h(dx), _h(dgamma), _h(dbeta), beta_dx))


def softmax_fwd(x, y, scale=1.0):
    _ck(_lib.nz_op_softmax_fwd(_h(x), _h(y), scale))


def softmax_bwd(dy, y, dx, scale=1.0):
    _ck(_lib.nz_op_softmax_bwd(_h(dy), _h(y), _h(dx), scale))


def adam_step(param, grad, master, m, v, lr, beta1=0.9, beta2=0.999, eps=1e-8,
              step=1, grad_scale=1.0, weight_decay=0.0):
    _ck(_lib.nz_op_adam_step(_h(param), _h(grad), _h(master), _h(m), _h(v), lr,
                             beta1, beta2, eps, step, grad_scale, weight_decay))


def sgd_step(param, grad, master, momentum, lr, mu=0.0, grad_scale=1.0):
    _ck(_lib.nz_op_sgd_step(_h(param), _h(grad), _h(master), _h(momentum), lr,
                            mu, grad_scale))


def dropout_fwd(x, y, mask, p, seed):
    _ck(_lib.nz_op_dropout_fwd(_h(x), _h(y), _h(mask), p, seed))


def dropout_bwd(dy, mask, dx, p):
    _ck(_lib.nz_op_dropout_bwd(_h(dy), _h(mask), _h(dx), p))


def mse_loss(y, target, loss, dy=None, grad_scale=1.0):
    _ck(_lib.nz_op_mse_loss(_h(y), _h(target), _h(loss), _h(dy), grad_scale))


def ce_loss(logits, labels, loss, dlogits=None, grad_scale=1.0):
    _ck(_lib.nz_op_ce_loss(_h(logits), _h(labels), _h(loss), _h(dlogits),
                           grad_scale))


def add_out(out, x, y):
    _ck(_lib.nz_op_add_out(_h(out), _h(x), _h(y)))


def add_inplace(dst, src, alpha=1.0):
    _ck(_lib.nz_op_add_inplace(_h(dst), _h(src), alpha))


def scale_inplace(dst, alpha):
    _ck(_lib.nz_op_scale_inplace(_h(dst), alpha))


def cast_copy(dst, src):
    _ck(_lib.nz_op_cast_copy(_h(dst), _h(src)))


def transpose2d(src, dst):
    _ck(_lib.nz_op_transpose2d(_h(src), _h(dst)))


def colsum(src, dst, beta=0.0):
    _ck(_lib.nz_op_colsum(_h(src), _h(dst), beta))


def reduce_sum(src, out):
    _ck(_lib.nz_op_reduce_sum(_h(src), _h(out)))


def l2norm_sq(src, out):
    _ck(_lib.nz_op_l2norm_sq(_h(src), _h(out)))


def attn_split_qkv(qkv, q, k, v, B, S, heads, dh):
    _ck(_lib.nz_op_attn_split_qkv(_h(qkv), _h(q), _h(k), _h(v), B, S, heads, dh))


def attn_fused_fwd(q, k, v, probs, ctx_out, batch_heads, S, dh, scale):
    _ck(_lib.nz_op_attn_fused_fwd(_h(q), _h(k), _h(v), _h(probs), _h(ctx_out),
                                  batch_heads, S, dh, scale))


def attn_fused_bwd(q, k, v, probs, dctx, dq, dk, dv, batch_heads, S, dh, scale):
    _ck(_lib.nz_op_attn_fused_bwd(_h(q), _h(k), _h(v), _h(probs), _h(dctx),
                                  _h(dq), _h(dk), _h(dv), batch_heads, S, dh,
                                  scale))


def attn_fwd_packed(qkv, probs, out, B, S, heads, dh, scale):
    _ck(_lib.nz_op_attn_fwd_packed(_h(qkv), _h(probs), _h(out), B, S, heads,
                                   dh, scale))


def attn_bwd_packed(qkv, probs, dmerged, dqkv, B, S, heads, dh, scale):
    _ck(_lib.nz_op_attn_bwd_packed(_h(qkv), _h(probs), _h(dmerged), _h(dqkv),
                                   B, S, heads, dh, scale))


def attn_merge(ctx_t, out, B, S, heads, dh):
    _ck(_lib.nz_op_attn_merge(_h(ctx_t), _h(out), B, S, heads, dh))


# ------------------------------------------------------------------ comm ---
class Comm:
    def __init__(self, handle):
        self._h = handle
        if not self._h:
            raise NezhaError(_lib.nz_last_error().decode())

    @staticmethod
    def uid_bytes():
        return _lib.nz_comm_uid_bytes()

    @staticmethod
    def get_uid():
        n = _lib.nz_comm_uid_bytes()
        buf = ctypes.create_string_buffer(n)
        _ck(_lib.nz_comm_get_uid(buf))
        return buf.raw

    @staticmethod
    def rccl(ctx, nranks, rank, uid):
        buf = ctypes.create_string_buffer(uid, len(uid))
        c = Comm(_lib.nz_comm_init_rccl(ctx._h, nranks, rank, buf))
        c._ctx = ctx
        return c

    @staticmethod
    def tcp(nranks, rank, host="127.0.0.1", port=29701):
        return Comm(_lib.nz_comm_init_tcp(nranks, rank, host.encode(), port))

    def __del__(self):
        if getattr(self, "_h", None) and _lib:
            _lib.nz_comm_free(self._h)
            self._h = None

    @property
    def rank(self):
        return _lib.nz_comm_rank(self._h)

    @property
    def nranks(self):
        return _lib.nz_comm_nranks(self._h)

    def barrier(self):
        _ck(_lib.nz_comm_barrier(self._h))

    def allreduce_scalar(self, v, op="sum"):
        buf = ctypes.c_float(v)
        _ck(_lib.nz_comm_allreduce_scalar(self._h, ctypes.byref(buf),
                                          1 if op == "max" else 0))
        return buf.value

    def allreduce_tensor(self, ctx, t, op="sum"):
        _ck(_lib.nz_comm_allreduce_tensor(self._h, ctx._h, _h(t),
                                          1 if op == "max" else 0))

    def broadcast_tensor(self, ctx, t, root=0):
        _ck(_lib.nz_comm_broadcast_tensor(self._h, ctx._h, _h(t), root))


# --------------------------------------------------------------- trainer ---
class Trainer:
    """Native trainer handle: model + optimizer + tape live in C++."""

    def __init__(self, handle):
        self._h = handle
        if not self._h:
            raise NezhaError(_lib.nz_last_error().decode())

    @staticmethod
    def mlp(ctx, layers=4, dim=4096, batch=8192, optim="adam", lr=1e-4,
            bucket_bytes=0, seed=1234):
        t = Trainer(_lib.nz_trainer_new_mlp(ctx._h, layers, dim, batch,
                                            optim.encode(), lr, bucket_bytes,
                                            seed))
        t._ctx = ctx  # keep the context alive past this trainer
        return t

    @staticmethod
    def bert(ctx, layers=12, hidden=768, ffn=3072, heads=12, seq=128, batch=32,
             optim="adam", lr=1e-4, bucket_bytes=0, seed=1234):
        t = Trainer(_lib.nz_trainer_new_bert(ctx._h, layers, hidden, ffn, heads,
                                             seq, batch, optim.encode(), lr,
                                             bucket_bytes, seed))
        t._ctx = ctx
        return t

    def __del__(self):
        if getattr(self, "_h", None) and _lib:
            _lib.nz_trainer_free(self._h)
            self._h = None

    @property
    def num_params(self):
        return _lib.nz_trainer_num_params(self._h)

    def attach_ddp(self, comm):
        _ck(_lib.nz_trainer_attach_ddp(self._h, comm._h))

    def reseed_data(self, seed):
        _ck(_lib.nz_trainer_reseed_data(self._h, seed))

    def step(self):
        _ck(_lib.nz_trainer_step(self._h))

    def loss(self):
        v = ctypes.c_float()
        _ck(_lib.nz_trainer_loss(self._h, ctypes.byref(v)))
        return v.value

    @property
    def step_count(self):
        return _lib.nz_trainer_step_count(self._h)

    @property
    def samples_per_step(self):
        return _lib.nz_trainer_samples_per_step(self._h)

    @property
    def flops_per_step(self):
        return _lib.nz_trainer_flops_per_step(self._h)

    def save(self, path):
        _ck(_lib.nz_ckpt_save(self._h, path.encode()))

    def save_local(self, path):
        """this rank's state, no rank gating/collective (test hook)"""
        _ck(_lib.nz_ckpt_save_local(self._h, path.encode()))

    def load(self, path):
        _ck(_lib.nz_ckpt_load(self._h, path.encode()))
