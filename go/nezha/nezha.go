//go:build cgo

// Package nezha is the public Go API of the nezha MI355X-native tensor +
// dense-NN training runtime (capability C1, SURVEY.md §0.3).
//
// It is a thin cgo mirror over the stable C ABI in include/nezha.h — the
// exact seam the engine exposes; all compute runs in the native C++/HIP
// engine (libnezha.so). This package contains no logic of its own so it
// cannot drift from the C ABI. The build environment of this repo has no
// Go toolchain (SURVEY.md §0.2), so this package is compile-verified only
// where Go is available:
//
//	CGO_CFLAGS="-I${REPO}/include" CGO_LDFLAGS="-L${REPO}/build -lnezha" go build ./go/nezha
package nezha

/*
#cgo CFLAGS: -I${SRCDIR}/../../include
#cgo LDFLAGS: -L${SRCDIR}/../../build -lnezha -Wl,-rpath,${SRCDIR}/../../build
#include <stdlib.h>
#include "nezha.h"
*/
import "C"

import (
	"errors"
	"runtime"
	"unsafe"
)

// DType of a tensor.
type DType int

const (
	F32  DType = C.NZ_F32
	BF16 DType = C.NZ_BF16
)

// Device kind.
type Device int

const (
	CPU Device = C.NZ_CPU
	HIP Device = C.NZ_HIP
)

func lastErr() error { return errors.New(C.GoString(C.nz_last_error())) }

func check(s C.nz_status) error {
	if s != 0 {
		return lastErr()
	}
	return nil
}

// Version returns the engine version string.
func Version() string { return C.GoString(C.nz_version()) }

// HipDeviceCount returns the number of visible MI355X GPUs.
func HipDeviceCount() int { return int(C.nz_hip_device_count()) }

// Ctx owns a device (allocator + streams).
type Ctx struct{ h *C.nz_ctx }

func NewCtx(dev Device, index int) (*Ctx, error) {
	h := C.nz_ctx_new(C.int(dev), C.int(index))
	if h == nil {
		return nil, lastErr()
	}
	c := &Ctx{h}
	runtime.SetFinalizer(c, func(c *Ctx) { C.nz_ctx_free(c.h) })
	return c, nil
}

func (c *Ctx) Sync() error { return check(C.nz_ctx_sync(c.h)) }

// Tensor is a dense row-major device tensor.
type Tensor struct {
	h     *C.nz_tensor
	Shape []int64
	DType DType
	ctx   *Ctx
}

func (c *Ctx) NewTensor(dtype DType, shape ...int64) (*Tensor, error) {
	dims := make([]C.int64_t, len(shape))
	for i, s := range shape {
		dims[i] = C.int64_t(s)
	}
	h := C.nz_tensor_new(c.h, C.int(dtype), C.int(len(shape)), &dims[0])
	if h == nil {
		return nil, lastErr()
	}
	t := &Tensor{h, append([]int64(nil), shape...), dtype, c}
	runtime.SetFinalizer(t, func(t *Tensor) { C.nz_tensor_free(t.h) })
	return t, nil
}

func (t *Tensor) Numel() int64 { return int64(C.nz_tensor_numel(t.h)) }

// Upload copies raw bytes (f32 or bf16 bit patterns) to the device.
func (t *Tensor) Upload(data []byte) error {
	return check(C.nz_tensor_upload(t.h, unsafe.Pointer(&data[0]), C.size_t(len(data))))
}

// Download copies the tensor's raw bytes back to the host.
func (t *Tensor) Download(out []byte) error {
	return check(C.nz_tensor_download(t.h, unsafe.Pointer(&out[0]), C.size_t(len(out))))
}

// UploadF32 uploads a []float32 into an F32 tensor.
func (t *Tensor) UploadF32(vals []float32) error {
	return check(C.nz_tensor_upload(t.h, unsafe.Pointer(&vals[0]), C.size_t(len(vals)*4)))
}

// DownloadF32 downloads an F32 tensor.
func (t *Tensor) DownloadF32(out []float32) error {
	return check(C.nz_tensor_download(t.h, unsafe.Pointer(&out[0]), C.size_t(len(out)*4)))
}

func (t *Tensor) Fill(v float32) error { return check(C.nz_tensor_fill(t.h, C.float(v))) }

func (t *Tensor) RandNormal(seed uint64, mean, std float32) error {
	return check(C.nz_tensor_rand_normal(t.h, C.uint64_t(seed), C.float(mean), C.float(std)))
}

func (t *Tensor) RandUniform(seed uint64, lo, hi float32) error {
	return check(C.nz_tensor_rand_uniform(t.h, C.uint64_t(seed), C.float(lo), C.float(hi)))
}

func th(t *Tensor) *C.nz_tensor {
	if t == nil {
		return nil
	}
	return t.h
}

// Gemm computes C = alpha*op(A).op(B) + beta*C with optional fused
// bias(+GELU) epilogue on the hand-written CDNA4 MFMA kernel (K1).
func Gemm(a, b, c *Tensor, transA, transB bool, alpha, beta float32, bias *Tensor, gelu bool, savePre *Tensor) error {
	ta, tb, g := C.int(0), C.int(0), C.int(0)
	if transA {
		ta = 1
	}
	if transB {
		tb = 1
	}
	if gelu {
		g = 1
	}
	return check(C.nz_op_gemm(a.h, b.h, c.h, ta, tb, C.float(alpha), C.float(beta), th(bias), g, th(savePre)))
}

// Softmax computes y = softmax(scale*x) row-wise (K4).
func Softmax(x, y *Tensor, scale float32) error {
	return check(C.nz_op_softmax_fwd(x.h, y.h, C.float(scale)))
}

// LayerNorm computes y = LN(x)*gamma+beta over the last dim (K3).
func LayerNorm(x, gamma, beta, y, mean, rstd *Tensor, eps float32) error {
	return check(C.nz_op_layernorm_fwd(x.h, gamma.h, beta.h, y.h, mean.h, rstd.h, C.float(eps)))
}

// AdamStep applies one fused Adam update on a flat bucket (K6).
func AdamStep(param, grad, master, m, v *Tensor, lr, beta1, beta2, eps float32, step int, gradScale, weightDecay float32) error {
	return check(C.nz_op_adam_step(param.h, grad.h, master.h, m.h, v.h,
		C.float(lr), C.float(beta1), C.float(beta2), C.float(eps), C.int(step),
		C.float(gradScale), C.float(weightDecay)))
}

// Comm is a communicator (RCCL over xGMI on GPU; TCP loopback for CPU tests).
type Comm struct{ h *C.nz_comm }

// CommUID returns a fresh RCCL unique id (rank 0 calls, others receive it).
func CommUID() ([]byte, error) {
	n := int(C.nz_comm_uid_bytes())
	buf := make([]byte, n)
	if err := check(C.nz_comm_get_uid(unsafe.Pointer(&buf[0]))); err != nil {
		return nil, err
	}
	return buf, nil
}

// NewRcclComm joins an RCCL communicator of nranks as rank.
func NewRcclComm(ctx *Ctx, nranks, rank int, uid []byte) (*Comm, error) {
	h := C.nz_comm_init_rccl(ctx.h, C.int(nranks), C.int(rank), unsafe.Pointer(&uid[0]))
	if h == nil {
		return nil, lastErr()
	}
	c := &Comm{h}
	runtime.SetFinalizer(c, func(c *Comm) { C.nz_comm_free(c.h) })
	return c, nil
}

func (c *Comm) Rank() int   { return int(C.nz_comm_rank(c.h)) }
func (c *Comm) NRanks() int { return int(C.nz_comm_nranks(c.h)) }
func (c *Comm) Barrier() error {
	return check(C.nz_comm_barrier(c.h))
}

// Trainer drives synthetic-data training of a built-in model (MLP / BERT
// encoder) entirely in the native engine: fwd -> loss -> bwd (-> bucketed
// RCCL all-reduce overlapped with backward) -> fused Adam/SGD step.
type Trainer struct {
	h   *C.nz_trainer
	ctx *Ctx
}

// NewMLPTrainer builds the C11 model: layers x [Linear dim->dim + bias + GELU].
func NewMLPTrainer(ctx *Ctx, layers, dim, batch int64, optim string, lr float32, bucketBytes int64, seed uint64) (*Trainer, error) {
	co := C.CString(optim)
	defer C.free(unsafe.Pointer(co))
	h := C.nz_trainer_new_mlp(ctx.h, C.int64_t(layers), C.int64_t(dim), C.int64_t(batch), co, C.float(lr), C.int64_t(bucketBytes), C.uint64_t(seed))
	if h == nil {
		return nil, lastErr()
	}
	t := &Trainer{h, ctx}
	runtime.SetFinalizer(t, func(t *Trainer) { C.nz_trainer_free(t.h) })
	return t, nil
}

// NewBertTrainer builds the C13 model: a BERT-base-class encoder stack.
func NewBertTrainer(ctx *Ctx, layers, hidden, ffn, heads, seq, batch int64, optim string, lr float32, bucketBytes int64, seed uint64) (*Trainer, error) {
	co := C.CString(optim)
	defer C.free(unsafe.Pointer(co))
	h := C.nz_trainer_new_bert(ctx.h, C.int64_t(layers), C.int64_t(hidden), C.int64_t(ffn), C.int64_t(heads), C.int64_t(seq), C.int64_t(batch), co, C.float(lr), C.int64_t(bucketBytes), C.uint64_t(seed))
	if h == nil {
		return nil, lastErr()
	}
	t := &Trainer{h, ctx}
	runtime.SetFinalizer(t, func(t *Trainer) { C.nz_trainer_free(t.h) })
	return t, nil
}

func (t *Trainer) NumParams() int64 { return int64(C.nz_trainer_num_params(t.h)) }

// AttachDDP enables data-parallel training over the communicator (C5).
func (t *Trainer) AttachDDP(c *Comm) error { return check(C.nz_trainer_attach_ddp(t.h, c.h)) }

// Step runs one training step (asynchronous on GPU).
func (t *Trainer) Step() error { return check(C.nz_trainer_step(t.h)) }

// Loss downloads the last step's loss (synchronizes).
func (t *Trainer) Loss() (float32, error) {
	var v C.float
	if err := check(C.nz_trainer_loss(t.h, &v)); err != nil {
		return 0, err
	}
	return float32(v), nil
}

// Save writes an .npz checkpoint (params + f32 master + Adam moments, C2).
func (t *Trainer) Save(path string) error {
	cp := C.CString(path)
	defer C.free(unsafe.Pointer(cp))
	return check(C.nz_ckpt_save(t.h, cp))
}

// SaveLocal writes THIS rank's state to its own file (no rank gating, no
// collective) — test hook for cross-rank agreement checks.
func (t *Trainer) SaveLocal(path string) error {
	cp := C.CString(path)
	defer C.free(unsafe.Pointer(cp))
	return check(C.nz_ckpt_save_local(t.h, cp))
}

// Load restores an .npz checkpoint and (under DDP) broadcasts rank 0's state.
func (t *Trainer) Load(path string) error {
	cp := C.CString(path)
	defer C.free(unsafe.Pointer(cp))
	return check(C.nz_ckpt_load(t.h, cp))
}
