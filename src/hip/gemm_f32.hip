// K1 (f32 path): exact-f32 GEMM on the CDNA4 f32-input MFMA
// (v_mfma_f32_16x16x4_f32 — guide §3: bitwise a k-ordered fmaf chain at the
// 157 TF f32 vector rate, ~2.4x an f32 VALU kernel). The bf16 kernels in
// gemm.hip are the training hot path; this completes the op surface for
// f32 GPU tensors (all four transpose variants + strided batch + epilogue).
// Structure: 128x128 tile, 4 waves (2x2, 64x64 each = 4x4 fragments of
// 16x16), BK=32 staged in padded LDS, double-buffered.
#include "src/hip/hip_common.h"
#include "src/hip/hip_ops.h"

namespace nz {
namespace hip {

using f32x4_ = __attribute__((ext_vector_type(4))) float;

namespace {

constexpr int FBM = 128, FBN = 128, FBK = 32;
constexpr int FSTRIDE = FBK + 4;  // f32 elems; 36*4=144 B rows, 16-B aligned
constexpr int FTHREADS = 256;

struct GemmF32Args {
  const float* A;
  const float* B;
  float* C;
  const float* bias;
  float* save_pre;
  int64_t M, N, K;
  int64_t lda, ldb;
  int64_t sA, sB, sC;
  float alpha, beta;
  int fuse_gelu;
};

// stage op(A) tile [FBM][FBK] rows-of-k into lds (+pad). float4 vector
// loads along the stored-contiguous dim; scatter for the transposed case.
template <bool TRANS>
__device__ void fstage_a(const float* __restrict__ A, float* __restrict__ lds,
                         int64_t m0, int64_t k0, int64_t M, int64_t lda) {
  constexpr int ELEMS = FBM * FBK / FTHREADS;  // 16 per thread as 4 x 4
#pragma unroll
  for (int rep = 0; rep < ELEMS / 4; ++rep) {
    int e = (rep * FTHREADS + (int)threadIdx.x) * 4;
    if (!TRANS) {
      int m = e / FBK, k = e % FBK;
      int64_t gm = m0 + m;
      f32x4_ v = {};
      if (gm < M) v = *(const f32x4_*)(A + gm * lda + k0 + k);
      *(f32x4_*)(lds + m * FSTRIDE + k) = v;
    } else {
      int k = e / FBM, m = e % FBM;
      int64_t gm = m0 + m;
      f32x4_ v = {};
      if (gm + 3 < M) {
        v = *(const f32x4_*)(A + (k0 + k) * lda + gm);
      } else {
#pragma unroll
        for (int i = 0; i < 4; ++i)
          v[i] = (gm + i < M) ? A[(k0 + k) * lda + gm + i] : 0.f;
      }
#pragma unroll
      for (int i = 0; i < 4; ++i) lds[(m + i) * FSTRIDE + k] = v[i];
    }
  }
}

template <bool TRANS>
__device__ void fstage_b(const float* __restrict__ B, float* __restrict__ lds,
                         int64_t n0, int64_t k0, int64_t N, int64_t ldb) {
  constexpr int ELEMS = FBN * FBK / FTHREADS;
#pragma unroll
  for (int rep = 0; rep < ELEMS / 4; ++rep) {
    int e = (rep * FTHREADS + (int)threadIdx.x) * 4;
    if (!TRANS) {  // stored [K,N]: contiguous n -> scatter into [n][k]
      int k = e / FBN, n = e % FBN;
      int64_t gn = n0 + n;
      f32x4_ v = {};
      if (gn + 3 < N) {
        v = *(const f32x4_*)(B + (k0 + k) * ldb + gn);
      } else {
#pragma unroll
        for (int i = 0; i < 4; ++i)
          v[i] = (gn + i < N) ? B[(k0 + k) * ldb + gn + i] : 0.f;
      }
#pragma unroll
      for (int i = 0; i < 4; ++i) lds[(n + i) * FSTRIDE + k] = v[i];
    } else {  // stored [N,K]: contiguous k -> direct rows
      int n = e / FBK, k = e % FBK;
      int64_t gn = n0 + n;
      f32x4_ v = {};
      if (gn < N) v = *(const f32x4_*)(B + gn * ldb + k0 + k);
      *(f32x4_*)(lds + n * FSTRIDE + k) = v;
    }
  }
}

template <bool TA, bool TB>
__launch_bounds__(FTHREADS) __global__ void gemm_f32_k(GemmF32Args args) {
  __shared__ float As[2][FBM * FSTRIDE];
  __shared__ float Bs[2][FBN * FSTRIDE];

  const int64_t batch = blockIdx.z;
  const float* A = args.A + batch * args.sA;
  const float* B = args.B + batch * args.sB;
  float* C = args.C + batch * args.sC;
  float* save_pre = args.save_pre ? args.save_pre + batch * args.sC : nullptr;

  const int64_t m0 = (int64_t)blockIdx.y * FBM;
  const int64_t n0 = (int64_t)blockIdx.x * FBN;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 1, wc = wid & 1;

  f32x4_ acc[4][4] = {};
  const int64_t ntiles = (args.K + FBK - 1) / FBK;

  fstage_a<TA>(A, As[0], m0, 0, args.M, args.lda);
  fstage_b<TB>(B, Bs[0], n0, 0, args.N, args.ldb);
  __syncthreads();

  int cur = 0;
  for (int64_t t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      fstage_a<TA>(A, As[cur ^ 1], m0, (t + 1) * FBK, args.M, args.lda);
      fstage_b<TB>(B, Bs[cur ^ 1], n0, (t + 1) * FBK, args.N, args.ldb);
    }
    const float* Ab = As[cur];
    const float* Bb = Bs[cur];
#pragma unroll
    for (int kk = 0; kk < FBK; kk += 4) {
      // v_mfma_f32_16x16x4_f32: lane l supplies A[l&15][k=l>>4], B[k][l&15]
      const int kidx = kk + (lane >> 4);
      float a_frag[4], b_frag[4];
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
        a_frag[fm] = Ab[(wr * 64 + fm * 16 + (lane & 15)) * FSTRIDE + kidx];
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
        b_frag[fn] = Bb[(wc * 64 + fn * 16 + (lane & 15)) * FSTRIDE + kidx];
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int fn = 0; fn < 4; ++fn)
#pragma unroll
      for (int rI = 0; rI < 4; ++rI) {
        int64_t grow = m0 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
        int64_t gcol = n0 + wc * 64 + fn * 16 + (lane & 15);
        if (grow >= args.M || gcol >= args.N) continue;
        int64_t idx = grow * args.N + gcol;
        float v = args.alpha * acc[fm][fn][rI];
        if (args.beta != 0.f) v += args.beta * C[idx];
        if (args.bias) v += args.bias[gcol];
        if (save_pre) save_pre[idx] = v;
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = v;
      }
}

void launch_f32(const GemmF32Args& args, bool tA, bool tB, int64_t batch,
                hipStream_t s) {
  NZ_CHECK(args.K % 4 == 0, "hip f32 gemm requires K % 4 == 0");
  dim3 grid((args.N + FBN - 1) / FBN, (args.M + FBM - 1) / FBM, batch);
  dim3 block(FTHREADS);
  if (!tA && !tB)
    gemm_f32_k<false, false><<<grid, block, 0, s>>>(args);
  else if (!tA && tB)
    gemm_f32_k<false, true><<<grid, block, 0, s>>>(args);
  else if (tA && !tB)
    gemm_f32_k<true, false><<<grid, block, 0, s>>>(args);
  else
    gemm_f32_k<true, true><<<grid, block, 0, s>>>(args);
  NZ_KERNEL_CHECK();
}

}  // namespace

void gemm_f32(const Tensor& A, const Tensor& B, Tensor& C, bool transA,
              bool transB, float alpha, float beta,
              const ops::GemmEpilogue* epi) {
  GemmF32Args args{};
  args.M = C.rows();
  args.N = C.cols();
  args.K = transA ? A.rows() : A.cols();
  args.A = (const float*)A.data();
  args.B = (const float*)B.data();
  args.C = (float*)C.data();
  args.lda = A.cols();
  args.ldb = B.cols();
  args.alpha = alpha;
  args.beta = beta;
  if (epi) {
    args.bias = epi->bias ? (const float*)epi->bias->data() : nullptr;
    args.save_pre = epi->save_pre ? (float*)epi->save_pre->data() : nullptr;
    args.fuse_gelu = epi->gelu ? 1 : 0;
  }
  launch_f32(args, transA, transB, 1, stream_of(C));
}

void gemm_f32_batched(const Tensor& A, const Tensor& B, Tensor& C, bool transA,
                      bool transB, float alpha, int64_t batch, int64_t M,
                      int64_t N, int64_t K, int64_t sA, int64_t sB, int64_t sC) {
  GemmF32Args args{};
  args.A = (const float*)A.data();
  args.B = (const float*)B.data();
  args.C = (float*)C.data();
  args.M = M;
  args.N = N;
  args.K = K;
  args.lda = transA ? M : K;
  args.ldb = transB ? K : N;
  args.sA = sA;
  args.sB = sB;
  args.sC = sC;
  args.alpha = alpha;
  launch_f32(args, transA, transB, batch, stream_of(C));
}

}  // namespace hip
}  // namespace nz
