// shifu_ops.hip — hand-written CDNA4 (gfx950 / MI355X) kernels for shifu_amd.
//
// Implements the MI355X-native replacements for the tensor ops the reference's
// TF graphs instantiate:
//   K1  fused dense GEMM + bias + activation forward (MFMA bf16, f32 accum)
//   K2  backward GEMMs: dX = dZ·Wᵀ (NT), dW = Xᵀ·dZ (TN, f32 out), db colsum,
//       fused activation-gradient
//   K3  fused sigmoid + weighted MSE / sigmoid-CE loss (+ gradient)
//   K4  fused Adam / Adadelta(TF semantics) / SGD / Adagrad over the flat
//       fp32 parameter arena
//   --  embedding arena gather + rowwise sparse updates (Wide&Deep/DeepFM)
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//   * wave64; blocks are 256 threads (4 waves)
//   * GEMM: 128x128 tile, BK=32, mfma_f32_16x16x32_bf16, LDS staged with
//     +16B row padding (conflict-free ds_read_b128 column groups), B held
//     transposed in LDS so B-fragments are contiguous 16B reads
//   * bf16 global loads vectorized (short4/short8-equivalent widths)
//   * elementwise/optimizer kernels are grid-stride, f32x4 / bf16x8 vectorized
// No CUDA compatibility paths; gfx950 only.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <vector>

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) short s16x4;   // 8B  = 4 bf16
typedef __attribute__((ext_vector_type(8))) short s16x8;   // 16B = 8 bf16

#define DEVINL __device__ __forceinline__

static inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_GPU(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")
#define CHECK_BF16(x) TORCH_CHECK((x).scalar_type() == at::kBFloat16, #x " must be bf16")
#define CHECK_F32(x) TORCH_CHECK((x).scalar_type() == at::kFloat, #x " must be fp32")

// ---------------------------------------------------------------------------
// activations (ids shared with shifu_amd/ops/linear.py)
// ---------------------------------------------------------------------------
enum Act { ACT_NONE = 0, ACT_SIGMOID = 1, ACT_TANH = 2, ACT_RELU = 3, ACT_LEAKY = 4 };
#define LEAKY_SLOPE 0.01f

DEVINL float act_fwd(float z, int act) {
  switch (act) {
    case ACT_SIGMOID: return 1.0f / (1.0f + __expf(-z));
    case ACT_TANH:    return tanhf(z);
    case ACT_RELU:    return z > 0.0f ? z : 0.0f;
    case ACT_LEAKY:   return z > 0.0f ? z : LEAKY_SLOPE * z;
    default:          return z;
  }
}

// activation gradient expressed through y = act(z) (valid for this act set)
DEVINL float act_grad_from_y(float y, int act) {
  switch (act) {
    case ACT_SIGMOID: return y * (1.0f - y);
    case ACT_TANH:    return 1.0f - y * y;
    case ACT_RELU:    return y > 0.0f ? 1.0f : 0.0f;
    case ACT_LEAKY:   return y > 0.0f ? 1.0f : LEAKY_SLOPE;
    default:          return 1.0f;
  }
}

// ---------------------------------------------------------------------------
// GEMM: C[M,N] = A_eff[M,K] * B_eff[K,N]  (bf16 in, f32 accumulate)
//   TA=0: A stored [M,K] (lda=K)   TA=1: A stored [K,M] (lda=M), A_eff=A^T
//   TB=0: B stored [K,N] (ldb=N)   TB=1: B stored [N,K] (ldb=K), B_eff=B^T
// Epilogue: EPI_PLAIN (bf16 out), EPI_BIAS_ACT (bf16 out, +bias, act),
//           EPI_F32 (f32 out — wgrad)
// One 128x128 tile per 256-thread block; per wave a 64x64 sub-tile =
// 4x4 fragments of 16x16; mfma_f32_16x16x32_bf16 over BK=32 K-steps.
// ---------------------------------------------------------------------------
enum Epi { EPI_PLAIN = 0, EPI_BIAS_ACT = 1, EPI_F32 = 2 };

#define BM 128
#define BN 128
#define BK 64
#define BKP (BK + 8)   // +16B row pad: 36-dword row stride; 16-lane b128 groups
                       // hit 16 distinct banks (gcd(36,64)=4, r<16 all distinct)

// SPLITK: blockIdx.z partitions the K (reduction) range; each part atomically
// accumulates into a zero-initialized f32 C.  Fills the 256-CU chip for
// wgrad shapes whose M/N tile grid alone is far below 256 workgroups.
template <int TA, int TB, int EPI, typename OUT_T, bool SPLITK = false>
__global__ __launch_bounds__(256)
void gemm_tile_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                      OUT_T* __restrict__ C, const bf16* __restrict__ bias,
                      int M, int N, int K, int act) {
  __shared__ bf16 As[BM][BKP];   // A_eff[m][k]
  __shared__ bf16 Bs[BN][BKP];   // B_eff[k][n] transposed: Bs[n][k]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;                    // 0..3
  const int wr = (wave >> 1) * 64;              // wave row offset in tile
  const int wc = (wave & 1) * 64;               // wave col offset in tile
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int r16 = lane & 15;        // fragment row/col within 16
  const int kgrp = lane >> 4;       // 0..3 -> k-offset group *8

  int k_lo = 0, k_hi = K;
  if (SPLITK) {
    int nz = gridDim.z;
    int chunk = ((K + nz - 1) / nz + BK - 1) / BK * BK;   // BK-aligned split
    k_lo = blockIdx.z * chunk;
    k_hi = min(K, k_lo + chunk);
    if (k_lo >= K) return;
  }

  for (int k0 = k_lo; k0 < k_hi; k0 += BK) {
    // ---- stage A tile: As[m][k] = A_eff[m0+m][k0+k] ----
    if (TA == 0) {
      // A[M,K]: 16B loads along k; 4 chunks/thread (BM*BK/8 = 1024 chunks)
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int i = tid + it * 256;
        int m = i >> 3;                 // /(BK/8)
        int kc = (i & 7) * 8;
        int gm = m0 + m, gk = k0 + kc;
        s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (gm < M) {
          const bf16* src = A + (long)gm * K + gk;
          if (gk + 8 <= K) v = *(const s16x8*)(src);
          else for (int j = 0; j < 8; ++j)
            ((short*)&v)[j] = (gk + j < K) ? ((const short*)src)[j] : (short)0;
        }
        *(s16x8*)&As[m][kc] = v;
      }
    } else {
      // A[K,M]: 16B loads along m (coalesced), scatter-transpose into LDS
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int i = tid + it * 256;
        int k = i >> 4;                 // /(BM/8)
        int mc = (i & 15) * 8;
        int gk = k0 + k, gm = m0 + mc;
        s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (gk < K) {
          const bf16* src = A + (long)gk * M + gm;
          if (gm + 8 <= M) v = *(const s16x8*)(src);
          else for (int j = 0; j < 8; ++j)
            ((short*)&v)[j] = (gm + j < M) ? ((const short*)src)[j] : (short)0;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) ((short*)&As[mc + j][k])[0] = ((short*)&v)[j];
      }
    }

    // ---- stage B tile: Bs[n][k] = B_eff[k0+k][n0+n] ----
    if (TB == 0) {
      // B[K,N]: 16B loads along n (coalesced), scatter-transpose
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int i = tid + it * 256;
        int k = i >> 4;
        int nc = (i & 15) * 8;
        int gk = k0 + k, gn = n0 + nc;
        s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (gk < K) {
          const bf16* src = B + (long)gk * N + gn;
          if (gn + 8 <= N) v = *(const s16x8*)(src);
          else for (int j = 0; j < 8; ++j)
            ((short*)&v)[j] = (gn + j < N) ? ((const short*)src)[j] : (short)0;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) ((short*)&Bs[nc + j][k])[0] = ((short*)&v)[j];
      }
    } else {
      // B[N,K]: rows are k-contiguous -> 16B vector LDS writes, no transpose
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int i = tid + it * 256;
        int n = i >> 3;
        int kc = (i & 7) * 8;
        int gn = n0 + n, gk = k0 + kc;
        s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (gn < N) {
          const bf16* src = B + (long)gn * K + gk;
          if (gk + 8 <= K) v = *(const s16x8*)(src);
          else for (int j = 0; j < 8; ++j)
            ((short*)&v)[j] = (gk + j < K) ? ((const short*)src)[j] : (short)0;
        }
        *(s16x8*)&Bs[n][kc] = v;
      }
    }

    __syncthreads();

    // ---- MFMA: 2 k-steps of K=32 per staged BK=64 tile ----
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int ko = ks * 32 + kgrp * 8;
#pragma unroll
      for (int fi = 0; fi < 4; ++fi) {
        bf16x8 a_frag = *(const bf16x8*)&As[wr + fi * 16 + r16][ko];
#pragma unroll
        for (int fj = 0; fj < 4; ++fj) {
          bf16x8 b_frag = *(const bf16x8*)&Bs[wc + fj * 16 + r16][ko];
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag, acc[fi][fj], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: D row = (lane>>4)*4 + r, col = lane&15 (per 16x16 frag) ----
  // bias hoisted: this lane's column set is 4 values, not 16
  float bvs[4] = {0.f, 0.f, 0.f, 0.f};
  if (EPI == EPI_BIAS_ACT) {
#pragma unroll
    for (int fj = 0; fj < 4; ++fj) {
      int col = n0 + wc + fj * 16 + r16;
      if (col < N) bvs[fj] = __bfloat162float(bias[col]);
    }
  }
#pragma unroll
  for (int fi = 0; fi < 4; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 4; ++fj) {
      int col = n0 + wc + fj * 16 + r16;
      if (col >= N) continue;
      float bv = bvs[fj];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr + fi * 16 + kgrp * 4 + r;
        if (row >= M) continue;
        float v = acc[fi][fj][r];
        if (EPI == EPI_BIAS_ACT) v = act_fwd(v + bv, act);
        if (EPI == EPI_F32) {
          if (SPLITK) atomicAdd(&((float*)C)[(long)row * N + col], v);
          else ((float*)C)[(long)row * N + col] = v;
        } else {
          ((bf16*)C)[(long)row * N + col] = __float2bfloat16(v);
        }
      }
    }
  }
}

template <int TA, int TB, int EPI, typename OUT_T>
static void launch_gemm(const bf16* A, const bf16* B, OUT_T* C, const bf16* bias,
                        long M, long N, long K, int act, hipStream_t s) {
  dim3 grid((N + BN - 1) / BN, (M + BM - 1) / BM);
  hipLaunchKernelGGL((gemm_tile_kernel<TA, TB, EPI, OUT_T, false>), grid, dim3(256), 0, s,
                     A, B, C, bias, (int)M, (int)N, (int)K, act);
}

// split-K variant (EPI_F32 wgrad): pick Z so the grid fills 256 CUs (>=512
// workgroups), each K-part >= one BK tile.  C must be zero-initialized.
template <int TA, int TB>
static void launch_gemm_splitk(const bf16* A, const bf16* B, float* C,
                               long M, long N, long K, hipStream_t s) {
  long gx = (N + BN - 1) / BN, gy = (M + BM - 1) / BM;
  long max_z = (K + BK - 1) / BK;
  long z = std::min<long>(std::max<long>(512 / std::max<long>(gx * gy, 1), 1), max_z);
  if (z <= 1) {
    launch_gemm<TA, TB, EPI_F32, float>(A, B, C, nullptr, M, N, K, 0, s);
    return;
  }
  dim3 grid((unsigned)gx, (unsigned)gy, (unsigned)z);
  hipLaunchKernelGGL((gemm_tile_kernel<TA, TB, EPI_F32, float, true>), grid, dim3(256), 0, s,
                     A, B, C, nullptr, (int)M, (int)N, (int)K, 0);
}

// ------------------------------ GEMM entry points ---------------------------
at::Tensor linear_act_fwd(at::Tensor x, at::Tensor w, at::Tensor b, long act) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  CHECK_GPU(w); CHECK_CONTIG(w); CHECK_BF16(w);
  CHECK_GPU(b); CHECK_CONTIG(b); CHECK_BF16(b);
  long Mb = x.size(0), K = x.size(1), N = w.size(1);
  TORCH_CHECK(w.size(0) == K, "shape mismatch x@w");
  auto y = at::empty({Mb, N}, x.options());
  launch_gemm<0, 0, EPI_BIAS_ACT, bf16>(
      (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(), (bf16*)y.data_ptr(),
      (const bf16*)b.data_ptr(), Mb, N, K, (int)act, cur_stream());
  return y;
}

at::Tensor gemm_nn_bf16(at::Tensor a, at::Tensor b) {
  CHECK_GPU(a); CHECK_CONTIG(a); CHECK_BF16(a);
  CHECK_GPU(b); CHECK_CONTIG(b); CHECK_BF16(b);
  long M = a.size(0), K = a.size(1), N = b.size(1);
  TORCH_CHECK(b.size(0) == K, "shape mismatch");
  auto c = at::empty({M, N}, a.options());
  launch_gemm<0, 0, EPI_PLAIN, bf16>((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                                     (bf16*)c.data_ptr(), nullptr, M, N, K, 0, cur_stream());
  return c;
}

// dx[B,K] = dz[B,N] @ w[K,N]^T
at::Tensor gemm_nt_bf16(at::Tensor dz, at::Tensor w) {
  CHECK_GPU(dz); CHECK_CONTIG(dz); CHECK_BF16(dz);
  CHECK_GPU(w); CHECK_CONTIG(w); CHECK_BF16(w);
  long Mb = dz.size(0), N = dz.size(1), K = w.size(0);
  TORCH_CHECK(w.size(1) == N, "shape mismatch dz@w^T");
  auto dx = at::empty({Mb, K}, dz.options());
  // C[M=Mb, N'=K] = dz[M,Kr=N] * (w^T)[Kr=N, N'=K]; w stored [K,N] => TB=1
  launch_gemm<0, 1, EPI_PLAIN, bf16>((const bf16*)dz.data_ptr(), (const bf16*)w.data_ptr(),
                                     (bf16*)dx.data_ptr(), nullptr, Mb, K, N, 0, cur_stream());
  return dx;
}

// dw[K,N] = x[B,K]^T @ dz[B,N]   (f32 out)
at::Tensor gemm_tn_f32(at::Tensor x, at::Tensor dz) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  CHECK_GPU(dz); CHECK_CONTIG(dz); CHECK_BF16(dz);
  long Bb = x.size(0), K = x.size(1), N = dz.size(1);
  TORCH_CHECK(dz.size(0) == Bb, "shape mismatch x^T@dz");
  auto dw = at::empty({K, N}, x.options().dtype(at::kFloat));
  hipMemsetAsync(dw.data_ptr(), 0, (size_t)K * N * 4, cur_stream());
  // C[M=K, N'=N] = (x^T)[M=K, Kr=B] * dz[Kr=B, N]; x stored [B,K] => TA=1
  launch_gemm_splitk<1, 0>((const bf16*)x.data_ptr(), (const bf16*)dz.data_ptr(),
                           (float*)dw.data_ptr(), K, N, Bb, cur_stream());
  return dw;
}

// ---------------------------------------------------------------------------
// v3 "NT" GEMM — THE fast path: both operands stored reduction-major.
//   C[M,N] = A[M,K] * B[N,K]^T   (bf16 in, f32 accumulate)
// Every hot GEMM routes here (fwd with weights stored [out,in]; dgrad with a
// per-step weight transpose; wgrad on transposed activations), so no
// scatter-transpose staging exists on the hot path.  Staging is
// global_load_lds (16 B/lane direct-to-LDS DMA, no VGPR round trip) into a
// LINEAR LDS image with the XOR swizzle applied on the per-lane SOURCE
// address and on the fragment reads (guide §5.4 rule 21); partial tiles fall
// back to a guarded ds_write path producing the identical swizzled image.
// Swizzle: granule(16B) index g at row r lives at g ^ (r&7) -> b128 fragment
// reads are <=2-way (rows r, r+8 share a bank), vs up-to-8-way unswizzled.
// ---------------------------------------------------------------------------
#define NT_BM 128
#define NT_BN 128
#define NT_BK 64
#define NT_LDS_BYTES (2 * NT_BM * NT_BK * 2)

typedef const __attribute__((address_space(1))) unsigned int* gas_ptr;
typedef __attribute__((address_space(3))) unsigned int* las_ptr;

// stage a 128x64 bf16 tile of P[RM rows, K cols] at (r0, k0) into lds
// (swizzled image).  Full tiles: 4 glds per wave; edges: guarded ds_writes.
DEVINL void nt_stage(const bf16* __restrict__ P, bf16* lds, int r0, int k0,
                     int RM, int K, int tid) {
  const int lane = tid & 63, wave = tid >> 6;
  if (r0 + NT_BM <= RM && k0 + NT_BK <= K) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int r = wave * 32 + c * 8 + (lane >> 3);
      int gsrc = (lane & 7) ^ (r & 7);
      const bf16* src = P + (long)(r0 + r) * K + k0 + gsrc * 8;
      bf16* dst = lds + (wave * 32 + c * 8) * NT_BK;   // wave-uniform; +lane*16B implicit
      __builtin_amdgcn_global_load_lds((gas_ptr)src, (las_ptr)dst, 16, 0, 0);
    }
  } else {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int i = tid + it * 256;
      int r = i >> 3, g = i & 7;
      int gr = r0 + r, gk = k0 + g * 8;
      s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (gr < RM) {
        const bf16* src = P + (long)gr * K + gk;
        if (gk + 8 <= K) v = *(const s16x8*)(src);
        else for (int j = 0; j < 8; ++j)
          ((short*)&v)[j] = (gk + j < K) ? ((const short*)src)[j] : (short)0;
      }
      *(s16x8*)(lds + r * NT_BK + ((g ^ (r & 7)) * 8)) = v;
    }
  }
}

// TRANSPOSED staging for the wgrad route: P is stored [K, R] (reduction-major
// ROWS — dz [B,M] / x [B,N] exactly as forward/act-grad produce them) and we
// build the SAME swizzled [128 x 64] image nt_stage does, so the MFMA loop's
// b128 fragment reads are untouched.  Global side: s16x8 along the contiguous
// R direction; a wave's 64 lanes read 64 consecutive K-rows at one 16 B
// column window, and the 8 windows sharing each 128 B line are issued by the
// same wave's other iterations (L1-resident after the first).  LDS side: b16
// scatter writes; bank = 4*((k>>3)^(r&7)) + ((k&7)>>1) depends only on (k,
// r&7), and each write instruction has r&7 uniform with 64 distinct k across
// the wave -> exactly 2-way conflicts (the two k sharing a dword).
// Alignment contract: R % 8 == 0 (callers guarantee; python falls back to
// the transpose route otherwise).
DEVINL void nt_stage_t(const bf16* __restrict__ P, bf16* lds, int r0, int k0,
                       int R, int K, int tid) {
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    int i = tid + it * 256;
    int k = i & 63;
    int mc = (i >> 6) * 8;
    int gk = k0 + k, gm = r0 + mc;
    s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (gk < K) {
      const bf16* src = P + (long)gk * R + gm;
      if (gm + 8 <= R) v = *(const s16x8*)(src);
      else for (int j = 0; j < 8; ++j)
        ((short*)&v)[j] = (gm + j < R) ? ((const short*)src)[j] : (short)0;
    }
    const int kh = k >> 3, kl = k & 7;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int r = mc + j;
      ((short*)(lds + r * NT_BK + ((kh ^ (r & 7)) << 3) + kl))[0] = ((short*)&v)[j];
    }
  }
}

// slab != 0 (split-K only): instead of atomicAdd into C[M,N], each z-part
// plain-stores its partial into C + blockIdx.z*slab (slab = M*N) and a
// separate splitk_reduce_kernel sums the parts — no f32 atomics, bitwise
// deterministic.  The host guarantees every z-part is non-empty in slab mode.
// TSTAGE=1: operands stored [K, M] / [K, N] (batch-major activations), staged
// by nt_stage_t — the transpose-free wgrad path.
template <int EPI, typename OUT_T, bool SPLITK = false, int TSTAGE = 0>
__global__ __launch_bounds__(256)
void gemm_nt_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                    OUT_T* __restrict__ C, const bf16* __restrict__ bias,
                    int M, int N, int K, int act, long slab) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* As = (bf16*)smem;                          // [128][64] swizzled
  bf16* Bs = (bf16*)(smem + NT_BM * NT_BK * 2);    // [128][64] swizzled

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int m0 = blockIdx.y * NT_BM;
  const int n0 = blockIdx.x * NT_BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int r16 = lane & 15;
  const int kgrp = lane >> 4;
  const int rx = r16 & 7;            // row&7 for every fragment row this lane reads

  int k_lo = 0, k_hi = K;
  if (SPLITK) {
    int nz = gridDim.z;
    int chunk = ((K + nz - 1) / nz + NT_BK - 1) / NT_BK * NT_BK;
    k_lo = blockIdx.z * chunk;
    k_hi = min(K, k_lo + chunk);
    if (k_lo >= K) return;
  }

  for (int k0 = k_lo; k0 < k_hi; k0 += NT_BK) {
    if (TSTAGE) {
      nt_stage_t(A, As, m0, k0, M, K, tid);
      nt_stage_t(B, Bs, n0, k0, N, K, tid);
    } else {
      nt_stage(A, As, m0, k0, M, K, tid);
      nt_stage(B, Bs, n0, k0, N, K, tid);
    }
    __syncthreads();   // hipcc emits the vmcnt(0) drain for in-flight glds here

#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int g = ks * 4 + kgrp;           // 16B granule index within the row
      const int go = (g ^ rx) * 8;           // swizzled element offset
#pragma unroll
      for (int fi = 0; fi < 4; ++fi) {
        bf16x8 a_frag = *(const bf16x8*)(As + (wr + fi * 16 + r16) * NT_BK + go);
#pragma unroll
        for (int fj = 0; fj < 4; ++fj) {
          bf16x8 b_frag = *(const bf16x8*)(Bs + (wc + fj * 16 + r16) * NT_BK + go);
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag, acc[fi][fj], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // bias hoisted: this lane's column set is 4 values, not 16
  float bvs[4] = {0.f, 0.f, 0.f, 0.f};
  if (EPI == EPI_BIAS_ACT) {
#pragma unroll
    for (int fj = 0; fj < 4; ++fj) {
      int col = n0 + wc + fj * 16 + r16;
      if (col < N) bvs[fj] = __bfloat162float(bias[col]);
    }
  }
#pragma unroll
  for (int fi = 0; fi < 4; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 4; ++fj) {
      int col = n0 + wc + fj * 16 + r16;
      if (col >= N) continue;
      float bv = bvs[fj];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr + fi * 16 + kgrp * 4 + r;
        if (row >= M) continue;
        float v = acc[fi][fj][r];
        if (EPI == EPI_BIAS_ACT) v = act_fwd(v + bv, act);
        if (EPI == EPI_F32) {
          if (SPLITK) {
            if (slab) ((float*)C)[(long)blockIdx.z * slab + (long)row * N + col] = v;
            else atomicAdd(&((float*)C)[(long)row * N + col], v);
          } else ((float*)C)[(long)row * N + col] = v;
        } else {
          ((bf16*)C)[(long)row * N + col] = __float2bfloat16(v);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// v4 "NT" GEMM — 256x256 tile, 8 waves, deep-pipelined glds ring.
// Same operand contract as v3 (A [M,K], B [N,K], both reduction-major) but:
//   * 512 threads = 8 waves as 2(M) x 4(N); per-wave output 128x64
//     (acc[8][4] f32x4 = 128 VGPRs); B fragments for a whole K-tile are
//     read once per tile and held in registers (32 VGPRs).
//   * 4 phases per K-tile; phase q computes accumulator rows 2q..2q+1
//     (16 MFMAs) after reading 4 A-fragments; one 16 KB half-tile of a
//     FUTURE K-tile is staged per phase by glds into a 2-deep ring
//     (A-halves one tile ahead, B-halves two tiles ahead — each region is
//     staged in the first phase after its previous tile's readers passed a
//     barrier, so placement-independent by construction).
//   * counted s_waitcnt vmcnt(4) at each K-tile boundary (never 0 in the
//     loop) + raw s_barrier per phase — loads stay in flight across
//     barriers (guide T3+T4); s_setprio(1) wraps each MFMA cluster (T5).
//   * K-edge tiles fall back to a guarded ds_write stage producing the
//     same swizzled image (granule g of row r lives at g ^ (r&7)).
// ---------------------------------------------------------------------------
#define V4_BM 256
#define V4_BN 256
#define V4_BK 64
// LDS regions (bytes): A ring buffer = tile parity; 32KB per operand-tile
#define V4_A(par) ((par) * 65536)
#define V4_B(par) ((par) * 65536 + 32768)

// T1: XCD-aware blockIdx remap (bijective for ANY block count, guide §5.5):
// the dispatcher places linear block b on XCD b%8, so remapping gives each
// XCD a CONTIGUOUS chunk of the row-major (by,bx) tile space — neighboring
// tiles that share A-row/B-col panels then hit the same per-XCD L2.
DEVINL void xcd_swizzle_xy(int& bx, int& by, int on) {
  if (!on) return;
  int gx = gridDim.x, gy = gridDim.y;
  int nwg = gx * gy;
  // Narrow grids only (the bench tower shapes: N/256 <= 8 column tiles):
  // there a per-XCD chunk of row-major tile space = many M-rows x all
  // N-cols, so the whole B panel set stays L2-resident per XCD (+3-8%
  // measured on fwd/dgrad shapes).  On wide square grids the same strips
  // span every B panel and LOSE to the dispatcher's round-robin (-6% @8k).
  if (gx > 8 || nwg < 16) return;
  int lin = by * gx + bx;
  int q = nwg >> 3, r = nwg & 7;
  int xcd = lin & 7, idx = lin >> 3;
  int lin2 = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  bx = lin2 % gx;
  by = lin2 / gx;
}

DEVINL void v4_stage_half_glds(const bf16* __restrict__ P, char* lds_region,
                               int R0, int half, int k0, int K, int tid) {
  // one 16KB half (rows half*128..+127 of the 256-row region), 2 glds/thread
  const int lane = tid & 63, wave = tid >> 6;
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    int byte = c * 8192 + wave * 1024 + lane * 16;
    int row = byte >> 7;               // within half
    int g = (byte >> 4) & 7;
    int gsrc = g ^ (row & 7);          // swizzle on the SOURCE address
    const bf16* src = P + (long)(R0 + half * 128 + row) * K + k0 + gsrc * 8;
    char* dst = lds_region + half * 16384 + c * 8192 + wave * 1024;
    __builtin_amdgcn_global_load_lds((gas_ptr)src, (las_ptr)dst, 16, 0, 0);
  }
}

DEVINL void v4_stage_half_guarded(const bf16* __restrict__ P, char* lds_region,
                                  int R0, int half, int k0, int RM, int K, int tid) {
  // same image via guarded loads + ds_writes (edge tiles); 2 chunks/thread
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    int byte = c * 8192 + (tid >> 6) * 1024 + (tid & 63) * 16;
    int row = byte >> 7;
    int g = (byte >> 4) & 7;
    int gsrc = g ^ (row & 7);
    int grow = R0 + half * 128 + row;
    long gk = (long)k0 + gsrc * 8;
    s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (grow < RM) {
      const bf16* src = P + (long)grow * K + gk;
      if (gk + 8 <= K) v = *(const s16x8*)src;
      else for (int j = 0; j < 8; ++j)
        ((short*)&v)[j] = (gk + j < K) ? ((const short*)src)[j] : (short)0;
    }
    *(s16x8*)(lds_region + half * 16384 + byte) = v;
  }
}

// stage one slot: slot 0/1 = A half0/1, slot 2/3 = B half0/1, of k-tile kt
DEVINL void v4_stage_slot(const bf16* A, const bf16* B, char* smem, int slot,
                          int kt, int m0, int n0, int M, int N, int K, int tid) {
  int k0 = kt * V4_BK;
  int par = kt & 1;
  bool full_k = (k0 + V4_BK <= K);
  if (slot < 2) {
    bool full = full_k && (m0 + V4_BM <= M);
    if (full) v4_stage_half_glds(A, smem + V4_A(par), m0, slot, k0, K, tid);
    else v4_stage_half_guarded(A, smem + V4_A(par), m0, slot, k0, M, K, tid);
  } else {
    bool full = full_k && (n0 + V4_BN <= N);
    if (full) v4_stage_half_glds(B, smem + V4_B(par), n0, slot - 2, k0, K, tid);
    else v4_stage_half_guarded(B, smem + V4_B(par), n0, slot - 2, k0, N, K, tid);
  }
}

// VAR 0: phase barrier after each of q0/q1 plus a separate boundary barrier
// VAR 1: q1's phase barrier merges into the boundary one (2 barriers/tile);
//        safe: every region staged right after the merged barrier had its
//        last reads completed by the lgkmcnt(0) preceding it
template <int EPI, typename OUT_T, bool SPLITK = false, int VAR = 0>
__global__ __launch_bounds__(512, 1)
void gemm_nt_v4_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                       OUT_T* __restrict__ C, const bf16* __restrict__ bias,
                       int M, int N, int K, int act, long slab, int swz) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;            // 0..1 -> M offset wm*128
  const int wn = wave & 3;             // 0..3 -> N offset wn*64
  int bx = blockIdx.x, by = blockIdx.y;
  xcd_swizzle_xy(bx, by, swz);         // T1: per-XCD L2 affinity
  const int m0 = by * V4_BM;
  const int n0 = bx * V4_BN;
  const int r16 = lane & 15;
  const int kgrp = lane >> 4;
  const int rx = r16 & 7;

  int kt_lo = 0, kt_hi = (K + V4_BK - 1) / V4_BK;
  if (SPLITK) {
    int nz = gridDim.z;
    int per = (kt_hi + nz - 1) / nz;
    kt_lo = blockIdx.z * per;
    kt_hi = min(kt_hi, kt_lo + per);
    if (kt_lo >= kt_hi) return;
  }
  const int n_kt = kt_hi - kt_lo;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // prologue: A(t0), B(t0), B(t1) fully; then enter the loop which stages
  // A(t+1) in phases 0-1 and B(t+2) in phases 1-2 of each tile t.
  v4_stage_slot(A, B, smem, 0, kt_lo, m0, n0, M, N, K, tid);
  v4_stage_slot(A, B, smem, 1, kt_lo, m0, n0, M, N, K, tid);
  v4_stage_slot(A, B, smem, 2, kt_lo, m0, n0, M, N, K, tid);
  v4_stage_slot(A, B, smem, 3, kt_lo, m0, n0, M, N, K, tid);
  if (n_kt > 1) {
    v4_stage_slot(A, B, smem, 2, kt_lo + 1, m0, n0, M, N, K, tid);
    v4_stage_slot(A, B, smem, 3, kt_lo + 1, m0, n0, M, N, K, tid);
  }
  {
    bool b1_glds = (n_kt > 1) && ((kt_lo + 1) * V4_BK + V4_BK <= K) &&
                   (n0 + V4_BN <= N);
    if (b1_glds) asm volatile("s_waitcnt vmcnt(4) lgkmcnt(0)" ::: "memory");
    else asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < n_kt; ++t) {
    const int kt = kt_lo + t;
    const int par = kt & 1;
    const char* Ar = smem + V4_A(par);
    const char* Br = smem + V4_B(par);

    // B fragments for the whole tile, held in registers
    bf16x8 bfr[4][2];
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        int row = wn * 64 + nf * 16 + r16;
        int g = ks * 4 + kgrp;
        bfr[nf][ks] = *(const bf16x8*)(Br + row * 128 + ((g ^ rx) * 16));
      }

#pragma unroll
    for (int q = 0; q < 2; ++q) {
      // A fragments for accumulator rows 4q..4q+3
      bf16x8 afr[4][2];
#pragma unroll
      for (int m2 = 0; m2 < 4; ++m2)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          int row = wm * 128 + (q * 4 + m2) * 16 + r16;
          int g = ks * 4 + kgrp;
          afr[m2][ks] = *(const bf16x8*)(Ar + row * 128 + ((g ^ rx) * 16));
        }
      // stage schedule: q0 -> A halves (t+1); q1 -> B halves (t+2)
      if (q == 0 && t + 1 < n_kt) {
        v4_stage_slot(A, B, smem, 0, kt + 1, m0, n0, M, N, K, tid);
        v4_stage_slot(A, B, smem, 1, kt + 1, m0, n0, M, N, K, tid);
      }
      if (q == 1 && t + 2 < n_kt) {
        v4_stage_slot(A, B, smem, 2, kt + 2, m0, n0, M, N, K, tid);
        v4_stage_slot(A, B, smem, 3, kt + 2, m0, n0, M, N, K, tid);
      }

      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m2 = 0; m2 < 4; ++m2)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int nf = 0; nf < 4; ++nf)
            acc[q * 4 + m2][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[m2][ks], bfr[nf][ks], acc[q * 4 + m2][nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // phase barrier: bound wave skew so the next phase's staging never
      // overwrites a region a lagging wave still reads.  lgkmcnt(0) flushes
      // this wave's ds_writes (edge-tile guarded staging) before the barrier.
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (!(VAR == 1 && q == 1)) __builtin_amdgcn_s_barrier();
    }
    // K-tile boundary.  The only loads allowed to stay in flight are the
    // B(t+2) slots (4 glds) — and only when they were actually staged by
    // glds; with a guarded/absent B(t+2), everything must drain or the
    // A(t+1) glds could outlive this wait and be read unlanded next tile.
    {
      bool b2_glds = (t + 2 < n_kt) && ((kt + 2) * V4_BK + V4_BK <= K) &&
                     (n0 + V4_BN <= N);
      if (b2_glds) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
  }

  // epilogue (bias hoisted: 4 loads per lane, not 32)
  float bvs[4] = {0.f, 0.f, 0.f, 0.f};
  if (EPI == EPI_BIAS_ACT) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int col = n0 + wn * 64 + nf * 16 + r16;
      if (col < N) bvs[nf] = __bfloat162float(bias[col]);
    }
  }
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int col = n0 + wn * 64 + nf * 16 + r16;
      if (col >= N) continue;
      float bv = bvs[nf];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wm * 128 + mf * 16 + kgrp * 4 + r;
        if (row >= M) continue;
        float v = acc[mf][nf][r];
        if (EPI == EPI_BIAS_ACT) v = act_fwd(v + bv, act);
        if (EPI == EPI_F32) {
          if (SPLITK) {
            if (slab) ((float*)C)[(long)blockIdx.z * slab + (long)row * N + col] = v;
            else atomicAdd(&((float*)C)[(long)row * N + col], v);
          } else ((float*)C)[(long)row * N + col] = v;
        } else {
          ((bf16*)C)[(long)row * N + col] = __float2bfloat16(v);
        }
      }
    }
  }
}

#define V4_LDS_BYTES 131072

// ---------------------------------------------------------------------------
// v5 "NT" GEMM — v4's 256x256 glds ring restructured to the guide's 8-phase
// sub-tile interleave (cdna_hip_programming.md §5 "256² 8-phase template",
// T3+T4+T5): four phases per K-tile, each phase
//   { ds_read one accumulator-row-pair's A fragments (q=0 also reads the
//     whole tile's B fragments) ∥ glds-stage ONE 16KB half-tile of a future
//     tile → raw s_barrier → s_waitcnt lgkmcnt(0) → s_setprio(1) →
//     16 MFMA (2 acc rows x 4 nf x 2 k-steps) → s_setprio(0) → raw barrier }
// vs v4's two 32-MFMA phases staging two half-tiles each.  The counted
// boundary wait (vmcnt(4), never 0 in the loop) is unchanged; the finer
// interleave is what lifts MfmaUtil (guide regime table: 8ph is the
// prerequisite for T2/T5 to pay).  Staging schedule and region-safety
// argument are v4's: A(t+1) halves in phases 0-1 (region's last reader was
// tile t-1), B(t+2) halves in phases 2-3 (B(t) fragments live in registers
// after phase 0's lgkmcnt(0)).
// The bf16 epilogue stages C tiles through LDS (free after the main loop:
// each wave owns 16KB = its 128x64 bf16 tile) so global stores are 16-byte
// dwordx4 instead of v4's 2-byte scalars.
// ---------------------------------------------------------------------------
// VAR 0: two barriers per phase (template-literal form)
// VAR 1: ONE barrier per phase — region safety still holds because every
//        stage target's last reader finished >= 2 barriers before the stage
//        issues (reads complete at the lgkmcnt(0) directly after their
//        phase's barrier, i.e. before the NEXT barrier)
template <int EPI, typename OUT_T, bool SPLITK = false, int VAR = 0>
__global__ __launch_bounds__(512, 1)
void gemm_nt_v5_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                       OUT_T* __restrict__ C, const bf16* __restrict__ bias,
                       int M, int N, int K, int act, long slab, int swz) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;            // 0..1 -> M offset wm*128
  const int wn = wave & 3;             // 0..3 -> N offset wn*64
  int bx = blockIdx.x, by = blockIdx.y;
  xcd_swizzle_xy(bx, by, swz);         // T1: per-XCD L2 affinity
  const int m0 = by * V4_BM;
  const int n0 = bx * V4_BN;
  const int r16 = lane & 15;
  const int kgrp = lane >> 4;
  const int rx = r16 & 7;

  int kt_lo = 0, kt_hi = (K + V4_BK - 1) / V4_BK;
  if (SPLITK) {
    int nz = gridDim.z;
    int per = (kt_hi + nz - 1) / nz;
    kt_lo = blockIdx.z * per;
    kt_hi = min(kt_hi, kt_lo + per);
    if (kt_lo >= kt_hi) return;
  }
  const int n_kt = kt_hi - kt_lo;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // prologue (v4's): tile t0 fully + B(t0+1); A(t0+1) arrives in phases 0-1.
  v4_stage_slot(A, B, smem, 0, kt_lo, m0, n0, M, N, K, tid);
  v4_stage_slot(A, B, smem, 1, kt_lo, m0, n0, M, N, K, tid);
  v4_stage_slot(A, B, smem, 2, kt_lo, m0, n0, M, N, K, tid);
  v4_stage_slot(A, B, smem, 3, kt_lo, m0, n0, M, N, K, tid);
  if (n_kt > 1) {
    v4_stage_slot(A, B, smem, 2, kt_lo + 1, m0, n0, M, N, K, tid);
    v4_stage_slot(A, B, smem, 3, kt_lo + 1, m0, n0, M, N, K, tid);
  }
  {
    bool b1_glds = (n_kt > 1) && ((kt_lo + 1) * V4_BK + V4_BK <= K) &&
                   (n0 + V4_BN <= N);
    if (b1_glds) asm volatile("s_waitcnt vmcnt(4) lgkmcnt(0)" ::: "memory");
    else asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < n_kt; ++t) {
    const int kt = kt_lo + t;
    const int par = kt & 1;
    const char* Ar = smem + V4_A(par);
    const char* Br = smem + V4_B(par);
    bf16x8 bfr[4][2];

#pragma unroll
    for (int q = 0; q < 4; ++q) {
      // ---- ds-load this phase's register subtile --------------------------
      if (q == 0) {
        // whole-tile B fragments (8 reads) — registers for all four phases
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) {
            int row = wn * 64 + nf * 16 + r16;
            int g = ks * 4 + kgrp;
            bfr[nf][ks] = *(const bf16x8*)(Br + row * 128 + ((g ^ rx) * 16));
          }
      }
      bf16x8 afr[2][2];  // this phase's two accumulator rows
#pragma unroll
      for (int m2 = 0; m2 < 2; ++m2)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          int row = wm * 128 + (q * 2 + m2) * 16 + r16;
          int g = ks * 4 + kgrp;
          afr[m2][ks] = *(const bf16x8*)(Ar + row * 128 + ((g ^ rx) * 16));
        }

      // ---- stage ONE half-tile of a future K-tile -------------------------
      // q0/q1: A(t+1) halves; q2/q3: B(t+2) halves (see region-safety above)
      if (q < 2) {
        if (t + 1 < n_kt)
          v4_stage_slot(A, B, smem, q, kt + 1, m0, n0, M, N, K, tid);
      } else {
        if (t + 2 < n_kt)
          v4_stage_slot(A, B, smem, q, kt + 2, m0, n0, M, N, K, tid);
      }
      if (q == 0)  // 12 ds_reads this phase: pace the LDS queue (guide §5)
        asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");

      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m2 = 0; m2 < 2; ++m2)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int nf = 0; nf < 4; ++nf)
            acc[q * 2 + m2][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[m2][ks], bfr[nf][ks], acc[q * 2 + m2][nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (q < 3) {
        if (VAR == 0) __builtin_amdgcn_s_barrier();
      } else {
        // K-tile boundary (merged with phase 3's closing barrier): only the
        // B(t+2) glds may stay in flight — and only when actually glds-staged
        bool b2_glds = (t + 2 < n_kt) && ((kt + 2) * V4_BK + V4_BK <= K) &&
                       (n0 + V4_BN <= N);
        if (b2_glds) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
      }
    }
  }

  if (EPI == EPI_F32) {
    // f32 outputs (wgrad/split-K): plain 4-byte stores as v4
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int col = n0 + wn * 64 + nf * 16 + r16;
        if (col >= N) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = m0 + wm * 128 + mf * 16 + kgrp * 4 + r;
          if (row >= M) continue;
          float v = acc[mf][nf][r];
          if (SPLITK) {
            if (slab) ((float*)C)[(long)blockIdx.z * slab + (long)row * N + col] = v;
            else atomicAdd(&((float*)C)[(long)row * N + col], v);
          } else ((float*)C)[(long)row * N + col] = v;
        }
      }
    return;
  }

  // bf16 epilogue: stage through LDS so the global stores are 16B dwordx4.
  // Wave w owns smem[w*16384 .. +16384) = its [128][64] bf16 C-tile.
  float bvs[4] = {0.f, 0.f, 0.f, 0.f};
  if (EPI == EPI_BIAS_ACT) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int col = n0 + wn * 64 + nf * 16 + r16;
      if (col < N) bvs[nf] = __bfloat162float(bias[col]);
    }
  }
  // all LDS is dead after the main loop's final barrier; per-wave regions
  // need no further synchronization
  // 16B-chunk XOR swizzle (chunk ^= bit2(row)*4): the two kgrp halves of a
  // wave write rows 4 apart, which land on the same banks at a 128B row
  // stride; the swizzle separates them (write conflicts -> none)
  short* cw = (short*)(smem + wave * 16384);
#pragma unroll
  for (int mf = 0; mf < 8; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = acc[mf][nf][r];
        if (EPI == EPI_BIAS_ACT) v = act_fwd(v + bvs[nf], act);
        bf16 h = __float2bfloat16(v);
        int row = mf * 16 + kgrp * 4 + r;
        int col = nf * 16 + r16;
        int sw = (((row >> 2) & 1) << 5);           // swap 16B chunks by 64B
        cw[row * 64 + (col ^ sw)] = *(short*)&h;
      }
    }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  // read back rows: 16 x ds_read_b128 per lane; lane covers rows
  // (i*64+lane)/8, 16B chunk (lane&7) -> stores land 8 rows x 128B per instr
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    int elt = i * 64 + lane;        // 16B-chunk index within the wave tile
    int row = elt >> 3;             // 8 chunks per 64-col row
    int c8 = (elt & 7) * 8;         // first col of this chunk
    int grow = m0 + wm * 128 + row;
    int gcol = n0 + wn * 64 + c8;
    int sw = (((row >> 2) & 1) << 5);
    s16x8 v = *(const s16x8*)(cw + row * 64 + (c8 ^ sw));
    if (grow < M) {
      if (gcol + 8 <= N) {
        *(s16x8*)((bf16*)C + (long)grow * N + gcol) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (gcol + j < N)
            ((short*)((bf16*)C + (long)grow * N))[gcol + j] = ((short*)&v)[j];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// v6 "NT" GEMM — v4's loop skeleton on mfma_f32_32x32x16_bf16: half the MFMA
// instructions at a higher per-shape µbench ceiling (2382 vs 2075 TF,
// cdna_hip_programming.md §3), same LDS image/staging/barrier structure.
// Fragment maps (verified by mfma_probe32):
//   A: lane holds A[frag_row = lane&31][k = (lane>>5)*8 + j]
//   B: lane holds B[frag_col = lane&31][same k]  (NT operand, row-major [N,K])
//   C: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// Per wave: 4(M) x 2(N) fragments of 32x32 -> acc[4][2] f32x16 (128 VGPR);
// per K-tile 32 MFMAs in two 16-MFMA phases.  bf16 epilogue LDS-staged as v5.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(16))) float f32x16v;

template <int EPI, typename OUT_T, bool SPLITK = false>
__global__ __launch_bounds__(512, 1)
void gemm_nt_v6_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                       OUT_T* __restrict__ C, const bf16* __restrict__ bias,
                       int M, int N, int K, int act, long slab, int swz) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  int bx = blockIdx.x, by = blockIdx.y;
  xcd_swizzle_xy(bx, by, swz);
  const int m0 = by * V4_BM;
  const int n0 = bx * V4_BN;
  const int r32 = lane & 31;
  const int kg = lane >> 5;
  const int rx = r32 & 7;

  int kt_lo = 0, kt_hi = (K + V4_BK - 1) / V4_BK;
  if (SPLITK) {
    int nz = gridDim.z;
    int per = (kt_hi + nz - 1) / nz;
    kt_lo = blockIdx.z * per;
    kt_hi = min(kt_hi, kt_lo + per);
    if (kt_lo >= kt_hi) return;
  }
  const int n_kt = kt_hi - kt_lo;

  f32x16v acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x16v{};

  v4_stage_slot(A, B, smem, 0, kt_lo, m0, n0, M, N, K, tid);
  v4_stage_slot(A, B, smem, 1, kt_lo, m0, n0, M, N, K, tid);
  v4_stage_slot(A, B, smem, 2, kt_lo, m0, n0, M, N, K, tid);
  v4_stage_slot(A, B, smem, 3, kt_lo, m0, n0, M, N, K, tid);
  if (n_kt > 1) {
    v4_stage_slot(A, B, smem, 2, kt_lo + 1, m0, n0, M, N, K, tid);
    v4_stage_slot(A, B, smem, 3, kt_lo + 1, m0, n0, M, N, K, tid);
  }
  {
    bool b1_glds = (n_kt > 1) && ((kt_lo + 1) * V4_BK + V4_BK <= K) &&
                   (n0 + V4_BN <= N);
    if (b1_glds) asm volatile("s_waitcnt vmcnt(4) lgkmcnt(0)" ::: "memory");
    else asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < n_kt; ++t) {
    const int kt = kt_lo + t;
    const int par = kt & 1;
    const char* Ar = smem + V4_A(par);
    const char* Br = smem + V4_B(par);

    // B fragments for the whole tile (2 nf x 4 ks)
    bf16x8 bfr[2][4];
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        int row = wn * 64 + nf * 32 + r32;
        int g = ks * 2 + kg;
        bfr[nf][ks] = *(const bf16x8*)(Br + row * 128 + ((g ^ rx) * 16));
      }

#pragma unroll
    for (int q = 0; q < 2; ++q) {
      bf16x8 afr[2][4];
#pragma unroll
      for (int m2 = 0; m2 < 2; ++m2)
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          int row = wm * 128 + (q * 2 + m2) * 32 + r32;
          int g = ks * 2 + kg;
          afr[m2][ks] = *(const bf16x8*)(Ar + row * 128 + ((g ^ rx) * 16));
        }
      if (q == 0 && t + 1 < n_kt) {
        v4_stage_slot(A, B, smem, 0, kt + 1, m0, n0, M, N, K, tid);
        v4_stage_slot(A, B, smem, 1, kt + 1, m0, n0, M, N, K, tid);
      }
      if (q == 1 && t + 2 < n_kt) {
        v4_stage_slot(A, B, smem, 2, kt + 2, m0, n0, M, N, K, tid);
        v4_stage_slot(A, B, smem, 3, kt + 2, m0, n0, M, N, K, tid);
      }

      __builtin_amdgcn_s_setprio(1);
      // ks OUTER: all 4 independent accumulators issue between two touches
      // of the same one (32x32 dependent-accumulator latency)
#pragma unroll
      for (int ks = 0; ks < 4; ++ks)
#pragma unroll
        for (int m2 = 0; m2 < 2; ++m2)
#pragma unroll
          for (int nf = 0; nf < 2; ++nf)
            acc[q * 2 + m2][nf] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                afr[m2][ks], bfr[nf][ks], acc[q * 2 + m2][nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
    {
      bool b2_glds = (t + 2 < n_kt) && ((kt + 2) * V4_BK + V4_BK <= K) &&
                     (n0 + V4_BN <= N);
      if (b2_glds) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
  }

  if (EPI == EPI_F32) {
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int col = n0 + wn * 64 + nf * 32 + r32;
        if (col >= N) continue;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int row = m0 + wm * 128 + mf * 32 + (r & 3) + 8 * (r >> 2) + 4 * kg;
          if (row >= M) continue;
          float v = acc[mf][nf][r];
          if (SPLITK) {
            if (slab) ((float*)C)[(long)blockIdx.z * slab + (long)row * N + col] = v;
            else atomicAdd(&((float*)C)[(long)row * N + col], v);
          } else ((float*)C)[(long)row * N + col] = v;
        }
      }
    return;
  }

  // bf16 epilogue, LDS-staged (v5's scheme; same bank swizzle: the two kg
  // halves write rows 4 apart)
  float bvs[2] = {0.f, 0.f};
  if (EPI == EPI_BIAS_ACT) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      int col = n0 + wn * 64 + nf * 32 + r32;
      if (col < N) bvs[nf] = __bfloat162float(bias[col]);
    }
  }
  short* cw = (short*)(smem + wave * 16384);
#pragma unroll
  for (int mf = 0; mf < 4; ++mf)
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float v = acc[mf][nf][r];
        if (EPI == EPI_BIAS_ACT) v = act_fwd(v + bvs[nf], act);
        bf16 h = __float2bfloat16(v);
        int row = mf * 32 + (r & 3) + 8 * (r >> 2) + 4 * kg;
        int col = nf * 32 + r32;
        int sw = (((row >> 2) & 1) << 5);
        cw[row * 64 + (col ^ sw)] = *(short*)&h;
      }
    }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    int elt = i * 64 + lane;
    int row = elt >> 3;
    int c8 = (elt & 7) * 8;
    int grow = m0 + wm * 128 + row;
    int gcol = n0 + wn * 64 + c8;
    int sw = (((row >> 2) & 1) << 5);
    s16x8 v = *(const s16x8*)(cw + row * 64 + (c8 ^ sw));
    if (grow < M) {
      if (gcol + 8 <= N) {
        *(s16x8*)((bf16*)C + (long)grow * N + gcol) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (gcol + j < N)
            ((short*)((bf16*)C + (long)grow * N))[gcol + j] = ((short*)&v)[j];
      }
    }
  }
}

// 256^2-tile GEMM implementation selector (SHIFU_GEMM_IMPL):
//   v4  — 2-phase x 32 MFMA, 3 barriers/tile (round-1 shipping kernel)
//   v4b — v4 with the q1 phase barrier merged into the boundary (2/tile)
//   v5  — 4-phase x 16 MFMA, 1 half-tile stage/phase, 2 barriers/phase,
//         LDS-staged dwordx4 epilogue
//   v5b — v5 with one barrier per phase
enum GemmImpl { GI_V4 = 0, GI_V4B = 1, GI_V5 = 2, GI_V5B = 3, GI_V6 = 4 };

static int gemm_impl() {
  static int v = [] {
    const char* e = getenv("SHIFU_GEMM_IMPL");
    if (!e) return (int)GI_V4;
    if (!strcmp(e, "v4")) return (int)GI_V4;
    if (!strcmp(e, "v4b")) return (int)GI_V4B;
    if (!strcmp(e, "v5")) return (int)GI_V5;
    if (!strcmp(e, "v5b")) return (int)GI_V5B;
    if (!strcmp(e, "v6")) return (int)GI_V6;
    return (int)GI_V4;
  }();
  return v;
}

// launch whichever 256^2 kernel the selector picks (typed, so the argument
// marshalling stays hipLaunchKernelGGL's)
static int xcd_swz_on() {
  static int v = [] {
    const char* e = getenv("SHIFU_XCD_SWZ");
    return e ? atoi(e) : 1;
  }();
  return v;
}

template <int EPI, typename OUT_T, bool SPLITK>
static void launch_256(dim3 grid, const bf16* A, const bf16* B, OUT_T* C,
                       const bf16* bias, int M, int N, int K, int act,
                       long slab, hipStream_t s) {
  const int swz = xcd_swz_on();
#define SHIFU_LAUNCH_256(KERN)                                              \
  do {                                                                      \
    hipFuncSetAttribute((const void*)(KERN),                                \
                        hipFuncAttributeMaxDynamicSharedMemorySize,         \
                        V4_LDS_BYTES);                                      \
    hipLaunchKernelGGL((KERN), grid, dim3(512), V4_LDS_BYTES, s,            \
                       A, B, C, bias, M, N, K, act, slab, swz);             \
  } while (0)
  switch (gemm_impl()) {
    case GI_V4B: SHIFU_LAUNCH_256((gemm_nt_v4_kernel<EPI, OUT_T, SPLITK, 1>)); break;
    case GI_V5:  SHIFU_LAUNCH_256((gemm_nt_v5_kernel<EPI, OUT_T, SPLITK, 0>)); break;
    case GI_V5B: SHIFU_LAUNCH_256((gemm_nt_v5_kernel<EPI, OUT_T, SPLITK, 1>)); break;
    case GI_V6:  SHIFU_LAUNCH_256((gemm_nt_v6_kernel<EPI, OUT_T, SPLITK>)); break;
    default:     SHIFU_LAUNCH_256((gemm_nt_v4_kernel<EPI, OUT_T, SPLITK, 0>)); break;
  }
#undef SHIFU_LAUNCH_256
}

template <int EPI, typename OUT_T>
static void launch_nt_v4(const bf16* A, const bf16* B, OUT_T* C, const bf16* bias,
                         long M, long N, long K, int act, hipStream_t s) {
  dim3 grid((N + V4_BN - 1) / V4_BN, (M + V4_BM - 1) / V4_BM);
  launch_256<EPI, OUT_T, false>(grid, A, B, C, bias, (int)M, (int)N, (int)K,
                                act, 0L, s);
}

template <int EPI, typename OUT_T>
static void launch_nt(const bf16* A, const bf16* B, OUT_T* C, const bf16* bias,
                      long M, long N, long K, int act, hipStream_t s) {
  dim3 grid((N + NT_BN - 1) / NT_BN, (M + NT_BM - 1) / NT_BM);
  hipLaunchKernelGGL((gemm_nt_kernel<EPI, OUT_T, false>), grid, dim3(256),
                     NT_LDS_BYTES, s, A, B, C, bias, (int)M, (int)N, (int)K, act, 0L);
}


static long splitk_target_blocks() {
  static long t = [] {
    const char* e = getenv("SHIFU_SPLITK_TARGET");
    return e ? atol(e) : 512L;
  }();
  return t;
}

// v4 runs 1 block/CU (128 KB LDS), so its split-K sweet spot is ~256 blocks
// (one per CU) — measured 188us vs 223us at 512 blocks on [1024,1864,32768].
// v3 (256 threads, 32 KB LDS) needs 2+ blocks/CU, hence the 512 above.
static long splitk_target_blocks_v4() {
  static long t = [] {
    const char* e = getenv("SHIFU_SPLITK_TARGET_V4");
    return e ? atol(e) : 256L;
  }();
  return t;
}

// C[i] (+)= sum_z W[z*MN + i] — the slab-mode reduction (f32x4 grid-stride).
// ACC=1 accumulates into C (wgrad written straight into the flat-grad view:
// no separate AccumulateGrad elementwise add per layer per step).
template <int ACC>
__global__ __launch_bounds__(256)
void splitk_reduce_kernel(const float* __restrict__ W, float* __restrict__ C,
                          long MN, int z) {
  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long nthr = (long)gridDim.x * blockDim.x;
  if ((MN & 3) == 0) {
    long nv = MN >> 2;
    for (long i = t0; i < nv; i += nthr) {
      f32x4 s = ((const f32x4*)W)[i];
      for (int zz = 1; zz < z; ++zz) s += ((const f32x4*)(W + (long)zz * MN))[i];
      if (ACC) s += ((const f32x4*)C)[i];
      ((f32x4*)C)[i] = s;
    }
  } else {
    for (long i = t0; i < MN; i += nthr) {
      float s = W[i];
      for (int zz = 1; zz < z; ++zz) s += W[(long)zz * MN + i];
      if (ACC) s += C[i];
      C[i] = s;
    }
  }
}

static void launch_splitk_reduce(const float* W, float* C, long MN, long z,
                                 hipStream_t s, bool accumulate = false) {
  long work = (MN & 3) == 0 ? MN >> 2 : MN;
  int blocks = (int)std::min<long>((work + 255) / 256, 2048);
  if (accumulate)
    hipLaunchKernelGGL(splitk_reduce_kernel<1>, dim3(blocks), dim3(256), 0, s,
                       W, C, MN, (int)z);
  else
    hipLaunchKernelGGL(splitk_reduce_kernel<0>, dim3(blocks), dim3(256), 0, s,
                       W, C, MN, (int)z);
}

// Split-K accumulation mode (default slab): each z-part plain-stores its
// partial into a [z, M*N] workspace and splitk_reduce_kernel sums it —
// bitwise deterministic and measured 11-22% faster than f32 atomicAdd on the
// bench wgrad shapes (atomics to the same lines serialize in L2; see
// profiles/r1_measurements.md).  SHIFU_SPLITK_SLAB=0 restores atomics.
static bool splitk_slab_mode() {
  static int v = [] {
    const char* e = getenv("SHIFU_SPLITK_SLAB");
    return e ? atoi(e) : 1;
  }();
  return v != 0;
}

// Shrink z until the kernels' ceil-chunking leaves no empty z-part (slab mode
// requires every part to store its slab; fixpoint of per=ceil(t/z), z=ceil(t/per)).
static long splitk_no_empty_z(long tiles, long z) {
  z = std::min(std::max(z, 1L), std::max(tiles, 1L));
  for (;;) {
    long per = (tiles + z - 1) / z;
    long z2 = (tiles + per - 1) / per;
    if (z2 == z) return z;
    z = z2;
  }
}

// Minimum K-tiles per split-K part before the v4 ring pipeline is allowed:
// its prologue stages 3 tiles, so short parts waste a large fraction of the
// loop in fill/drain (measured: slower than v3 at reduction 8192 with
// 21-tile parts, faster at 32768 with >=24).
static long splitk_min_kt() {
  static long t = [] {
    const char* e = getenv("SHIFU_SPLITK_MINKT");
    return e ? atol(e) : 24L;
  }();
  return t;
}

// v4 split-K into c[M,N] f32 (handles its own zero/workspace; c may be
// uninit unless accumulate=true, in which case c's content is ADDED to).
static void run_nt_v4_splitk_f32(const bf16* A, const bf16* B, at::Tensor& c,
                                 long M, long N, long K, hipStream_t s, long z,
                                 bool accumulate = false) {
  dim3 grid((unsigned)((N + V4_BN - 1) / V4_BN),
            (unsigned)((M + V4_BM - 1) / V4_BM), (unsigned)z);
  float* cp = (float*)c.data_ptr();
  if (splitk_slab_mode()) {
    long kt = (K + V4_BK - 1) / V4_BK;
    long zs = splitk_no_empty_z(kt, z);
    grid.z = (unsigned)zs;
    auto w = at::empty({zs, M * N}, c.options());
    launch_256<EPI_F32, float, true>(grid, A, B, (float*)w.data_ptr(), nullptr,
                                     (int)M, (int)N, (int)K, 0, M * N, s);
    launch_splitk_reduce((const float*)w.data_ptr(), cp, M * N, zs, s, accumulate);
  } else {
    if (!accumulate) hipMemsetAsync(cp, 0, (size_t)M * N * 4, s);
    launch_256<EPI_F32, float, true>(grid, A, B, cp, nullptr,
                                     (int)M, (int)N, (int)K, 0, 0L, s);
  }
}

// v3 split-K into c[M,N] f32 (handles its own zero/workspace; c may be uninit).
static void run_nt_splitk_f32(const bf16* A, const bf16* B, at::Tensor& c,
                              long M, long N, long K, hipStream_t s,
                              bool accumulate = false) {
  long gx = (N + NT_BN - 1) / NT_BN, gy = (M + NT_BM - 1) / NT_BM;
  long max_z = (K + NT_BK - 1) / NT_BK;
  long z = std::min<long>(
      std::max<long>(splitk_target_blocks() / std::max<long>(gx * gy, 1), 1), max_z);
  // Prefer parts >=16 K-tiles when that still leaves >=256 blocks (measured:
  // [256,512,32768] 37.6us at z=32/256 blocks vs 43.6us at z=64/512 blocks;
  // shapes where the cap would under-fill the chip keep the 512-block z).
  // SHIFU_SPLITK_V3MINKT=1 disables the cap.
  static long v3minkt = [] {
    const char* e = getenv("SHIFU_SPLITK_V3MINKT");
    return e ? atol(e) : 16L;
  }();
  // Only for deep reductions (>=256 K-tiles): at K=8192 the same cap cost 5%
  // end-to-end on DeepFM (parts got too few blocks-in-flight per period).
  long z16 = std::max<long>(max_z / std::max(v3minkt, 1L), 1);
  if (max_z >= 256 && gx * gy * z16 >= 256) z = std::min(z, z16);
  float* cp = (float*)c.data_ptr();
  if (z <= 1) {
    if (accumulate) {
      auto tmp = at::empty({M, N}, c.options());
      launch_nt<EPI_F32, float>(A, B, (float*)tmp.data_ptr(), nullptr,
                                M, N, K, 0, s);
      c.add_(tmp.view_as(c));
    } else {
      launch_nt<EPI_F32, float>(A, B, cp, nullptr, M, N, K, 0, s);
    }
    return;
  }
  if (splitk_slab_mode()) {
    z = splitk_no_empty_z(max_z, z);
    auto w = at::empty({z, M * N}, c.options());
    hipLaunchKernelGGL((gemm_nt_kernel<EPI_F32, float, true>),
                       dim3((unsigned)gx, (unsigned)gy, (unsigned)z), dim3(256),
                       NT_LDS_BYTES, s, A, B, (float*)w.data_ptr(), nullptr,
                       (int)M, (int)N, (int)K, 0, M * N);
    launch_splitk_reduce((const float*)w.data_ptr(), cp, M * N, z, s, accumulate);
  } else {
    if (!accumulate) hipMemsetAsync(cp, 0, (size_t)M * N * 4, s);
    hipLaunchKernelGGL((gemm_nt_kernel<EPI_F32, float, true>),
                       dim3((unsigned)gx, (unsigned)gy, (unsigned)z), dim3(256),
                       NT_LDS_BYTES, s, A, B, cp, nullptr,
                       (int)M, (int)N, (int)K, 0, 0L);
  }
}

// v4 (256^2 pipelined) pays when both tile dims fill 256 rows and the K loop
// is deep enough to amortize the ring prologue.
static inline bool use_v4(long M, long N, long K) {
  static int disable = [] {
    const char* e = getenv("SHIFU_DISABLE_V4");
    return e ? atoi(e) : 0;
  }();
  if (disable) return false;
  // v4 runs 1 block/CU (128 KB LDS); below ~200 blocks it leaves CUs idle
  // and the 128^2-tile v3 (2-3 blocks/CU) wins.
  long blocks = ((M + V4_BM - 1) / V4_BM) * ((N + V4_BN - 1) / V4_BN);
  return M >= 512 && N >= 256 && K >= 256 && blocks >= 200;
}

// y[B,N] = act(x[B,K] @ w[N,K]^T + b)   — the hot forward (weights [out,in])
at::Tensor linear_nt_fwd(at::Tensor x, at::Tensor w, at::Tensor b, long act) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  CHECK_GPU(w); CHECK_CONTIG(w); CHECK_BF16(w);
  CHECK_GPU(b); CHECK_CONTIG(b); CHECK_BF16(b);
  long Mb = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "shape mismatch x@w^T");
  auto y = at::empty({Mb, N}, x.options());
  if (use_v4(Mb, N, K))
    launch_nt_v4<EPI_BIAS_ACT, bf16>((const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
                                     (bf16*)y.data_ptr(), (const bf16*)b.data_ptr(),
                                     Mb, N, K, (int)act, cur_stream());
  else
    launch_nt<EPI_BIAS_ACT, bf16>((const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
                                  (bf16*)y.data_ptr(), (const bf16*)b.data_ptr(),
                                  Mb, N, K, (int)act, cur_stream());
  return y;
}

// c[M,N] = a[M,K] @ b[N,K]^T (plain bf16 out)
at::Tensor gemm_ntv3_bf16(at::Tensor a, at::Tensor b) {
  CHECK_GPU(a); CHECK_CONTIG(a); CHECK_BF16(a);
  CHECK_GPU(b); CHECK_CONTIG(b); CHECK_BF16(b);
  long M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "shape mismatch a@b^T");
  auto c = at::empty({M, N}, a.options());
  if (use_v4(M, N, K))
    launch_nt_v4<EPI_PLAIN, bf16>((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                                  (bf16*)c.data_ptr(), nullptr, M, N, K, 0, cur_stream());
  else
    launch_nt<EPI_PLAIN, bf16>((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                               (bf16*)c.data_ptr(), nullptr, M, N, K, 0, cur_stream());
  return c;
}

// wgrad ACCUMULATED into `out` (the [M,N] f32 flat-grad view): out += a@b^T.
// Saves the fresh-dw allocation and the per-layer AccumulateGrad add.
void gemm_ntv3_f32_into(at::Tensor a, at::Tensor b, at::Tensor out) {
  CHECK_GPU(a); CHECK_CONTIG(a); CHECK_BF16(a);
  CHECK_GPU(b); CHECK_CONTIG(b); CHECK_BF16(b);
  CHECK_GPU(out); CHECK_CONTIG(out); CHECK_F32(out);
  long M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K && out.size(0) == M && out.size(1) == N,
              "shape mismatch out += a@b^T");
  long gx4 = (N + V4_BN - 1) / V4_BN, gy4 = (M + V4_BM - 1) / V4_BM;
  long kt4 = (K + V4_BK - 1) / V4_BK;
  long z24 = std::min<long>(
      std::max<long>(splitk_target_blocks_v4() / std::max<long>(gx4 * gy4, 1), 1),
      std::max<long>(kt4 / splitk_min_kt(), 1));
  if (M >= 256 && N >= 256 && z24 > 1 && gx4 * gy4 * z24 >= 200) {
    run_nt_v4_splitk_f32((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                         out, M, N, K, cur_stream(), z24, /*accumulate=*/true);
  } else {
    run_nt_splitk_f32((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                      out, M, N, K, cur_stream(), /*accumulate=*/true);
  }
}

// c[M,N] f32 = a[M,K] @ b[N,K]^T with split-K (wgrad)
at::Tensor gemm_ntv3_f32(at::Tensor a, at::Tensor b) {
  CHECK_GPU(a); CHECK_CONTIG(a); CHECK_BF16(a);
  CHECK_GPU(b); CHECK_CONTIG(b); CHECK_BF16(b);
  long M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "shape mismatch a@b^T");
  auto c = at::empty({M, N}, a.options().dtype(at::kFloat));
  long gx4 = (N + V4_BN - 1) / V4_BN, gy4 = (M + V4_BM - 1) / V4_BM;
  long kt4 = (K + V4_BK - 1) / V4_BK;
  // v4 where its grid fills the chip; v4 split-K only when each part keeps
  // >=splitk_min_kt() K-tiles (see splitk_min_kt above).  Each route covers
  // every in-range element of c, so no zero-fill is needed here.
  long z24 = std::min<long>(
      std::max<long>(splitk_target_blocks_v4() / std::max<long>(gx4 * gy4, 1), 1),
      std::max<long>(kt4 / splitk_min_kt(), 1));
  if (use_v4(M, N, K) && gx4 * gy4 >= 200) {
    launch_nt_v4<EPI_F32, float>((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                                 (float*)c.data_ptr(), nullptr, M, N, K, 0, cur_stream());
  } else if (M >= 256 && N >= 256 && z24 > 1 && gx4 * gy4 * z24 >= 200) {
    run_nt_v4_splitk_f32((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                         c, M, N, K, cur_stream(), z24);
  } else {
    run_nt_splitk_f32((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                      c, M, N, K, cur_stream());
  }
  return c;
}

// ---------------------------------------------------------------------------
// TT wgrad on the v3 skeleton: C[M,N] f32 = sum_b A[b,m] * B[b,n] with NO
// pre-transposes — nt_stage_t scatter-builds the identical swizzled image,
// the MFMA loop and the deterministic slab split-K are shared with v3.
// Replaces transpose_bf16(x) + gemm_ntv3_f32(dzT, xT): 1/3 the HBM traffic
// on the x operand (no transpose read+write) and two fewer launches per
// layer per step.
// ---------------------------------------------------------------------------
static void run_tt_splitk_f32(const bf16* A, const bf16* B, at::Tensor& c,
                              long M, long N, long K, hipStream_t s,
                              bool accumulate = false) {
  long gx = (N + NT_BN - 1) / NT_BN, gy = (M + NT_BM - 1) / NT_BM;
  long max_z = (K + NT_BK - 1) / NT_BK;
  long z = std::min<long>(
      std::max<long>(splitk_target_blocks() / std::max<long>(gx * gy, 1), 1), max_z);
  static long v3minkt = [] {
    const char* e = getenv("SHIFU_SPLITK_V3MINKT");
    return e ? atol(e) : 16L;
  }();
  long z16 = std::max<long>(max_z / std::max(v3minkt, 1L), 1);
  if (max_z >= 256 && gx * gy * z16 >= 256) z = std::min(z, z16);
  float* cp = (float*)c.data_ptr();
  if (z <= 1) {
    dim3 grid((unsigned)gx, (unsigned)gy);
    if (accumulate) {
      auto tmp = at::empty({M, N}, c.options());
      hipLaunchKernelGGL((gemm_nt_kernel<EPI_F32, float, false, 1>), grid,
                         dim3(256), NT_LDS_BYTES, s, A, B,
                         (float*)tmp.data_ptr(), nullptr,
                         (int)M, (int)N, (int)K, 0, 0L);
      c.add_(tmp.view_as(c));
    } else {
      hipLaunchKernelGGL((gemm_nt_kernel<EPI_F32, float, false, 1>), grid,
                         dim3(256), NT_LDS_BYTES, s, A, B, cp, nullptr,
                         (int)M, (int)N, (int)K, 0, 0L);
    }
    return;
  }
  if (splitk_slab_mode()) {
    z = splitk_no_empty_z(max_z, z);
    auto w = at::empty({z, M * N}, c.options());
    hipLaunchKernelGGL((gemm_nt_kernel<EPI_F32, float, true, 1>),
                       dim3((unsigned)gx, (unsigned)gy, (unsigned)z), dim3(256),
                       NT_LDS_BYTES, s, A, B, (float*)w.data_ptr(), nullptr,
                       (int)M, (int)N, (int)K, 0, M * N);
    launch_splitk_reduce((const float*)w.data_ptr(), cp, M * N, z, s, accumulate);
  } else {
    if (!accumulate) hipMemsetAsync(cp, 0, (size_t)M * N * 4, s);
    hipLaunchKernelGGL((gemm_nt_kernel<EPI_F32, float, true, 1>),
                       dim3((unsigned)gx, (unsigned)gy, (unsigned)z), dim3(256),
                       NT_LDS_BYTES, s, A, B, cp, nullptr,
                       (int)M, (int)N, (int)K, 0, 0L);
  }
}

// out[M,N] += a[R,M]^T @ b[R,N] — transpose-free wgrad straight into the
// flat-grad view.  Requires M % 8 == 0 && N % 8 == 0 (16 B row alignment for
// the staging loads; linear.py falls back to the transpose route otherwise).
void gemm_ttv3_f32_into(at::Tensor a, at::Tensor b, at::Tensor out) {
  CHECK_GPU(a); CHECK_CONTIG(a); CHECK_BF16(a);
  CHECK_GPU(b); CHECK_CONTIG(b); CHECK_BF16(b);
  CHECK_GPU(out); CHECK_CONTIG(out); CHECK_F32(out);
  long R = a.size(0), M = a.size(1), N = b.size(1);
  TORCH_CHECK(b.size(0) == R && out.size(0) == M && out.size(1) == N,
              "shape mismatch out += a^T@b");
  TORCH_CHECK((M & 7) == 0 && (N & 7) == 0, "ttv3 needs M,N %% 8 == 0");
  run_tt_splitk_f32((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                    out, M, N, R, cur_stream(), /*accumulate=*/true);
}

// c[M,N] f32 = a[R,M]^T @ b[R,N] (transpose-free wgrad, fresh output)
at::Tensor gemm_ttv3_f32(at::Tensor a, at::Tensor b) {
  CHECK_GPU(a); CHECK_CONTIG(a); CHECK_BF16(a);
  CHECK_GPU(b); CHECK_CONTIG(b); CHECK_BF16(b);
  long R = a.size(0), M = a.size(1), N = b.size(1);
  TORCH_CHECK(b.size(0) == R, "shape mismatch a^T@b");
  TORCH_CHECK((M & 7) == 0 && (N & 7) == 0, "ttv3 needs M,N %% 8 == 0");
  auto c = at::empty({M, N}, a.options().dtype(at::kFloat));
  run_tt_splitk_f32((const bf16*)a.data_ptr(), (const bf16*)b.data_ptr(),
                    c, M, N, R, cur_stream());
  return c;
}

// ---------------------------------------------------------------------------
// v5 "TT" wgrad GEMM — C[M,N] f32 = sum_b dz[b,m] * x[b,n], both operands
// stored reduction-major-ROWS ([B, M] / [B, N]) exactly as the activations
// come out of forward/act-grad: NO pre-transposes.  The column reads MFMA
// needs come from gfx950's hardware transpose-read (ds_read_b64_tr_b16):
// each image is stored in [4(b) x 16(col)] row-major blocks (glds fills the
// blocked layout straight from coalescing-friendly 16 B global pieces), and
// a tr read hands each lane column (l&15) of block (base + (l>>4)) — i.e.
// 4 b-values for its fragment column.  Block order within a 16-col group is
// f(bblk) = (bblk&1)*8 + bblk/2 so one read pair covers b = 8G..8G+7.
// Split-K over B via f32 atomics (zero-initialized C).
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(4))) short s16x4t;
typedef __attribute__((__vector_size__(4 * sizeof(short)))) short __attribute__((address_space(3)))* lds_v4s;

#define V5_BM 128
#define V5_BN 128
#define V5_BB 64    // reduction rows per staged tile
#define V5_IMG_BYTES 16384

// image element index e (0..8191) -> (b_local, col) of the logical [64][128]
DEVINL void v5_inverse(int e, int& b_local, int& col) {
  int sb = e >> 6;
  int nblk = sb >> 4;
  int fb = sb & 15;
  int bblk = (fb < 8) ? 2 * fb : 2 * (fb - 8) + 1;
  int j = (e >> 4) & 3;
  int c16 = e & 15;
  b_local = bblk * 4 + j;
  col = nblk * 16 + c16;
}

// stage one [64 x 128] blocked image; interior: 4 glds/thread
DEVINL void v5_stage(const bf16* __restrict__ P, char* img, int b0, int c0,
                     int R, int ld, int tid, bool interior) {
  const int lane = tid & 63, wave = tid >> 6;
  if (interior) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e0 = (tid + c * 256) * 8;
      int bl, col;
      v5_inverse(e0, bl, col);
      const bf16* src = P + (long)(b0 + bl) * ld + c0 + col;
      char* dst = img + wave * 1024 + c * 4096;
      __builtin_amdgcn_global_load_lds((gas_ptr)src, (las_ptr)dst, 16, 0, 0);
    }
  } else {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e0 = (tid + c * 256) * 8;
      int bl, col;
      v5_inverse(e0, bl, col);
      s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      long gb = b0 + bl;
      if (gb < R) {
        const bf16* src = P + gb * ld + c0 + col;
        for (int q = 0; q < 8; ++q)
          if (c0 + col + q < ld) ((short*)&v)[q] = ((const short*)src)[q];
      }
      *(s16x8*)(img + (long)e0 * 2) = v;
    }
  }
}

template <bool SPLITK>
__global__ __launch_bounds__(256)
void gemm_tt_kernel(const bf16* __restrict__ DZ, const bf16* __restrict__ X,
                    float* __restrict__ C, int M, int N, int R) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* Ai = smem;                      // buffer0: dz image (cols = m)
  char* Bi = smem + V5_IMG_BYTES;       // buffer0: x image  (cols = n)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int m0 = blockIdx.y * V5_BM;
  const int n0 = blockIdx.x * V5_BN;
  const int r16 = lane & 15;
  const int kgrp = lane >> 4;

  int b_lo = 0, b_hi = R;
  if (SPLITK) {
    int nz = gridDim.z;
    int chunk = ((R + nz - 1) / nz + V5_BB - 1) / V5_BB * V5_BB;
    b_lo = blockIdx.z * chunk;
    b_hi = min(R, b_lo + chunk);
    if (b_lo >= b_hi) return;
  }

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const bool a_int_mn = (m0 + V5_BM <= M);
  const bool b_int_mn = (n0 + V5_BN <= N);
  // tr-read base block offsets per (ks, readhalf): f(2G)=G basis
  const int roff[2][2] = {{0, 8}, {4, 12}};

  // double-buffered: stage tile t+1 while computing tile t; ONE counted
  // drain + raw barrier per tile (a __syncthreads per tile would stall on
  // the in-flight glds before every compute phase)
  {
    bool fb = (b_lo + V5_BB <= R);
    v5_stage(DZ, Ai, b_lo, m0, R, M, tid, fb && a_int_mn);
    v5_stage(X, Bi, b_lo, n0, R, N, tid, fb && b_int_mn);
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }
  int cur = 0;
  for (int b0 = b_lo; b0 < b_hi; b0 += V5_BB) {
    char* Ac = smem + cur * 2 * V5_IMG_BYTES;
    char* Bc = Ac + V5_IMG_BYTES;
    int bn = b0 + V5_BB;
    if (bn < b_hi) {
      char* An = smem + (cur ^ 1) * 2 * V5_IMG_BYTES;
      bool fb = (bn + V5_BB <= R);
      v5_stage(DZ, An, bn, m0, R, M, tid, fb && a_int_mn);
      v5_stage(X, An + V5_IMG_BYTES, bn, n0, R, N, tid, fb && b_int_mn);
    }
    const char* Ai2 = Ac;
    const char* Bi2 = Bc;

#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
      for (int fi = 0; fi < 4; ++fi) {
        const int nblkA = (wr + fi * 16) >> 4;
        bf16x8 afrag;
        {
          s16x4t r1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (lds_v4s)(Ai2 + (nblkA * 16 + roff[ks][0]) * 128 + lane * 8));
          s16x4t r2 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (lds_v4s)(Ai2 + (nblkA * 16 + roff[ks][1]) * 128 + lane * 8));
#pragma unroll
          for (int q = 0; q < 4; ++q) {
            ((short*)&afrag)[q] = r1[q];
            ((short*)&afrag)[q + 4] = r2[q];
          }
        }
#pragma unroll
        for (int fj = 0; fj < 4; ++fj) {
          const int nblkB = (wc + fj * 16) >> 4;
          bf16x8 bfrag;
          s16x4t r1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (lds_v4s)(Bi2 + (nblkB * 16 + roff[ks][0]) * 128 + lane * 8));
          s16x4t r2 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (lds_v4s)(Bi2 + (nblkB * 16 + roff[ks][1]) * 128 + lane * 8));
#pragma unroll
          for (int q = 0; q < 4; ++q) {
            ((short*)&bfrag)[q] = r1[q];
            ((short*)&bfrag)[q + 4] = r2[q];
          }
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, bfrag, acc[fi][fj], 0, 0, 0);
        }
      }
    }
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }

#pragma unroll
  for (int fi = 0; fi < 4; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 4; ++fj) {
      int coln = n0 + wc + fj * 16 + r16;
      if (coln >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int rowm = m0 + wr + fi * 16 + kgrp * 4 + r;
        if (rowm >= M) continue;
        if (SPLITK) atomicAdd(&C[(long)rowm * N + coln], acc[fi][fj][r]);
        else C[(long)rowm * N + coln] = acc[fi][fj][r];
      }
    }
  }
}

// dw[M=out, N=in] = dz[B, M]^T @ x[B, N]  — transpose-free wgrad
at::Tensor gemm_tt_f32(at::Tensor dz, at::Tensor x) {
  CHECK_GPU(dz); CHECK_CONTIG(dz); CHECK_BF16(dz);
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  long R = dz.size(0), M = dz.size(1), N = x.size(1);
  TORCH_CHECK(x.size(0) == R, "reduction mismatch dz^T@x");
  auto c = at::empty({M, N}, x.options().dtype(at::kFloat));
  hipMemsetAsync(c.data_ptr(), 0, (size_t)M * N * 4, cur_stream());
  long gx = (N + V5_BN - 1) / V5_BN, gy = (M + V5_BM - 1) / V5_BM;
  long max_z = (R + V5_BB - 1) / V5_BB;
  long z = std::min<long>(
      std::max<long>(splitk_target_blocks() / std::max<long>(gx * gy, 1), 1), max_z);
  auto s = cur_stream();
  if (z <= 1) {
    hipLaunchKernelGGL((gemm_tt_kernel<false>), dim3((unsigned)gx, (unsigned)gy),
                       dim3(256), 4 * V5_IMG_BYTES, s,
                       (const bf16*)dz.data_ptr(), (const bf16*)x.data_ptr(),
                       (float*)c.data_ptr(), (int)M, (int)N, (int)R);
  } else {
    hipLaunchKernelGGL((gemm_tt_kernel<true>), dim3((unsigned)gx, (unsigned)gy, (unsigned)z),
                       dim3(256), 4 * V5_IMG_BYTES, s,
                       (const bf16*)dz.data_ptr(), (const bf16*)x.data_ptr(),
                       (float*)c.data_ptr(), (int)M, (int)N, (int)R);
  }
  return c;
}

// ---------------------------------------------------------------------------
// tiled bf16 transpose: out[C,R] = in[R,C]^T.  64x64 LDS tiles (+8 pad),
// 16B coalesced global loads AND stores; scalar traffic stays inside LDS.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256)
void transpose_bf16_kernel(const bf16* __restrict__ in, bf16* __restrict__ out,
                           int R, int C) {
  // 66-short rows = 33-dword (odd) stride: the transposed column reads then
  // walk 33r mod 32 = distinct banks for all 64 rows (the old +8 pad made a
  // 36-dword stride -> 8-way conflicts).  Writes go as 4-byte pairs (any
  // wider LDS write would be misaligned at a 132-byte stride).
  __shared__ short tile[64 * 66];
  int rt = blockIdx.y * 64, ct = blockIdx.x * 64;
  int tid = threadIdx.x;
  // load 64x64: 512 chunks of 8, 2 per thread
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    int i = tid + it * 256;
    int r = i >> 3, g = (i & 7) * 8;
    int gr = rt + r, gc = ct + g;
    s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (gr < R) {
      const bf16* src = in + (long)gr * C + gc;
      if (gc + 8 <= C) v = *(const s16x8*)src;
      else for (int j = 0; j < 8; ++j)
        ((short*)&v)[j] = (gc + j < C) ? ((const short*)src)[j] : (short)0;
    }
#pragma unroll
    for (int j = 0; j < 8; j += 2)
      *(unsigned*)&tile[r * 66 + g + j] = *(unsigned*)&((short*)&v)[j];
  }
  __syncthreads();
  // store 64x64 transposed: thread writes out[ct+c][rt+r..r+7]
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    int i = tid + it * 256;
    int c = i >> 3, g = (i & 7) * 8;
    int oc = ct + c, orr = rt + g;
    if (oc >= C) continue;
    s16x8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j) ((short*)&v)[j] = tile[(g + j) * 66 + c];
    bf16* dst = out + (long)oc * R + orr;
    if (orr + 8 <= R) *(s16x8*)dst = v;
    else for (int j = 0; j < 8 && orr + j < R; ++j) ((short*)dst)[j] = ((short*)&v)[j];
  }
}

at::Tensor transpose_bf16(at::Tensor t) {
  CHECK_GPU(t); CHECK_CONTIG(t); CHECK_BF16(t);
  long R = t.size(0), C = t.size(1);
  auto out = at::empty({C, R}, t.options());
  dim3 grid((unsigned)((C + 63) / 64), (unsigned)((R + 63) / 64));
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, cur_stream(),
                     (const bf16*)t.data_ptr(), (bf16*)out.data_ptr(), (int)R, (int)C);
  return out;
}


// single-MFMA probe: d[16,16] = a[16,32] @ b[32,16] — fragment-mapping unit test
__global__ void mfma_probe_kernel(const bf16* a, const bf16* b, float* d) {
  int lane = threadIdx.x & 63;
  int r16 = lane & 15, kgrp = lane >> 4;
  bf16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    ((__bf16*)&af)[j] = *(const __bf16*)&a[r16 * 32 + kgrp * 8 + j];
    ((__bf16*)&bf)[j] = *(const __bf16*)&b[(kgrp * 8 + j) * 16 + r16];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) d[(kgrp * 4 + r) * 16 + r16] = acc[r];
}

at::Tensor mfma_probe(at::Tensor a, at::Tensor b) {
  CHECK_GPU(a); CHECK_BF16(a); CHECK_GPU(b); CHECK_BF16(b);
  auto d = at::zeros({16, 16}, a.options().dtype(at::kFloat));
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     (const bf16*)a.contiguous().data_ptr(),
                     (const bf16*)b.contiguous().data_ptr(), (float*)d.data_ptr());
  return d;
}

// 32x32x16 layout probe: a [32,16] row-major, b [32,16] row-major (NT — row
// n holds that column's k values); d[32,32] = a @ b^T.  Assumed maps:
//   A frag: lane holds A[lane&31][(lane>>5)*8 + j]
//   B frag: lane holds B^T[(lane>>5)*8 + j][lane&31] = bsrc[lane&31][...]
//   C/D   : col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
typedef __attribute__((ext_vector_type(16))) float f32x16;

__global__ void mfma_probe32_kernel(const bf16* a, const bf16* b, float* d) {
  int lane = threadIdx.x & 63;
  int r32 = lane & 31, kg = lane >> 5;
  bf16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    ((__bf16*)&af)[j] = *(const __bf16*)&a[r32 * 16 + kg * 8 + j];
    ((__bf16*)&bf)[j] = *(const __bf16*)&b[r32 * 16 + kg * 8 + j];
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    d[((r & 3) + 8 * (r >> 2) + 4 * kg) * 32 + r32] = acc[r];
}

at::Tensor mfma_probe32(at::Tensor a, at::Tensor b) {
  CHECK_GPU(a); CHECK_BF16(a); CHECK_GPU(b); CHECK_BF16(b);
  auto d = at::zeros({32, 32}, a.options().dtype(at::kFloat));
  hipLaunchKernelGGL(mfma_probe32_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     (const bf16*)a.contiguous().data_ptr(),
                     (const bf16*)b.contiguous().data_ptr(), (float*)d.data_ptr());
  return d;
}

// ---------------------------------------------------------------------------
// elementwise: fused activation gradient  dz = dy * act'(y)
// ---------------------------------------------------------------------------
__global__ void act_grad_kernel(const bf16* __restrict__ dy, const bf16* __restrict__ y,
                                bf16* __restrict__ dz, long n, int act) {
  long i = (long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  // vectorized by 8 (16B loads)
  long n8 = n / 8;
  for (long v = i; v < n8; v += stride) {
    s16x8 dyv = ((const s16x8*)dy)[v];
    s16x8 yv = ((const s16x8*)y)[v];
    s16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float dyf = __bfloat162float(((const bf16*)&dyv)[j]);
      float yf = __bfloat162float(((const bf16*)&yv)[j]);
      ((bf16*)&out)[j] = __float2bfloat16(dyf * act_grad_from_y(yf, act));
    }
    ((s16x8*)dz)[v] = out;
  }
  for (long j = n8 * 8 + i; j < n; j += stride) {
    float dyf = __bfloat162float(dy[j]);
    float yf = __bfloat162float(y[j]);
    dz[j] = __float2bfloat16(dyf * act_grad_from_y(yf, act));
  }
}

at::Tensor act_grad(at::Tensor dy, at::Tensor y, long act) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_BF16(dy);
  CHECK_GPU(y); CHECK_CONTIG(y); CHECK_BF16(y);
  auto dz = at::empty_like(dy);
  long n = dy.numel();
  int blocks = (int)std::min((n + 2047) / 2048 + 1, (long)2048);
  hipLaunchKernelGGL(act_grad_kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                     (const bf16*)dy.data_ptr(), (const bf16*)y.data_ptr(),
                     (bf16*)dz.data_ptr(), n, (int)act);
  return dz;
}

// ---------------------------------------------------------------------------
// colsum: db[N] = sum_b dz[b][n]  (f32 out)
// thread t of block bx owns column bx*256+t; row reads are coalesced.
// ---------------------------------------------------------------------------
// 2D grid: blockIdx.y picks a row chunk; partials atomicAdd'd — fills the
// 256-CU chip for any (B,N) (the 1D version left 255/256 CUs idle).
__global__ void colsum_kernel(const bf16* __restrict__ dz, float* __restrict__ db,
                              long B, long N, long rows_per_chunk) {
  long n = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= N) return;
  long b0 = (long)blockIdx.y * rows_per_chunk;
  long b1 = min(B, b0 + rows_per_chunk);
  float acc = 0.0f;
  for (long b = b0; b < b1; ++b) acc += __bfloat162float(dz[b * N + n]);
  if (gridDim.y == 1) db[n] = acc;
  else atomicAdd(&db[n], acc);
}

static void colsum_grid(long B, long N, long& gx, long& chunks, long& rows_per_chunk) {
  gx = (N + 255) / 256;
  chunks = std::min<long>(std::max<long>(1024 / std::max<long>(gx, 1), 1),
                          std::max<long>(B / 64, 1));
  rows_per_chunk = (B + chunks - 1) / chunks;
}

at::Tensor colsum_f32(at::Tensor dz) {
  CHECK_GPU(dz); CHECK_CONTIG(dz); CHECK_BF16(dz);
  long B = dz.size(0), N = dz.size(1);
  auto db = at::zeros({N}, dz.options().dtype(at::kFloat));
  long gx, chunks, rpc;
  colsum_grid(B, N, gx, chunks, rpc);
  hipLaunchKernelGGL(colsum_kernel, dim3((unsigned)gx, (unsigned)chunks), dim3(256),
                     0, cur_stream(),
                     (const bf16*)dz.data_ptr(), (float*)db.data_ptr(), B, N, rpc);
  return db;
}

// fused act-grad + bias-grad: dz = dy*act'(y) AND db = colsum(dz) in ONE pass
// over [B,N] — saves the full dz re-read a separate colsum would cost.
// one thread per column: row reads are coalesced across threads (2 KB per
// wave-row); measured 2 TB/s — the 8-col vectorized variant loses to
// per-column atomic contention at this geometry.
__global__ void act_grad_colsum_kernel(const bf16* __restrict__ dy,
                                       const bf16* __restrict__ y,
                                       bf16* __restrict__ dz, float* __restrict__ db,
                                       long B, long N, long rows_per_chunk, int act,
                                       int accum) {
  long n = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= N) return;
  long b0 = (long)blockIdx.y * rows_per_chunk;
  long b1 = min(B, b0 + rows_per_chunk);
  float acc = 0.0f;
  for (long b = b0; b < b1; ++b) {
    long i = b * N + n;
    float g = __bfloat162float(dy[i]) * act_grad_from_y(__bfloat162float(y[i]), act);
    dz[i] = __float2bfloat16(g);
    acc += g;
  }
  if (!accum && gridDim.y == 1) db[n] = acc;
  else if (accum && gridDim.y == 1) db[n] += acc;
  else atomicAdd(&db[n], acc);
}

std::vector<at::Tensor> act_grad_colsum(at::Tensor dy, at::Tensor y, long act) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_BF16(dy);
  CHECK_GPU(y); CHECK_CONTIG(y); CHECK_BF16(y);
  long B = dy.size(0), N = dy.size(1);
  auto dz = at::empty_like(dy);
  auto db = at::zeros({N}, dy.options().dtype(at::kFloat));
  long gx, chunks, rpc;
  colsum_grid(B, N, gx, chunks, rpc);
  hipLaunchKernelGGL(act_grad_colsum_kernel, dim3((unsigned)gx, (unsigned)chunks),
                     dim3(256), 0, cur_stream(),
                     (const bf16*)dy.data_ptr(), (const bf16*)y.data_ptr(),
                     (bf16*)dz.data_ptr(), (float*)db.data_ptr(),
                     B, N, rpc, (int)act, 0);
  return {dz, db};
}

// act-grad with the colsum ACCUMULATED into db (the bias flat-grad view):
// the dzT-free partner of the ttv3 wgrad route — no transposed copy of dz
// is materialized at all (gemm_ttv3_f32_into reads dz batch-major).
at::Tensor act_grad_colsum_into(at::Tensor dy, at::Tensor y, long act,
                                at::Tensor db) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_BF16(dy);
  CHECK_GPU(y); CHECK_CONTIG(y); CHECK_BF16(y);
  CHECK_GPU(db); CHECK_F32(db);
  long B = dy.size(0), N = dy.size(1);
  TORCH_CHECK(db.numel() == N && db.is_contiguous(), "db view mismatch");
  auto dz = at::empty_like(dy);
  long gx, chunks, rpc;
  colsum_grid(B, N, gx, chunks, rpc);
  hipLaunchKernelGGL(act_grad_colsum_kernel, dim3((unsigned)gx, (unsigned)chunks),
                     dim3(256), 0, cur_stream(),
                     (const bf16*)dy.data_ptr(), (const bf16*)y.data_ptr(),
                     (bf16*)dz.data_ptr(), (float*)db.data_ptr(),
                     B, N, rpc, (int)act, 1);
  return dz;
}

// act-grad + colsum + TRANSPOSE in one pass: dz = dy*act'(y) written both
// row-major (dgrad operand) and transposed (wgrad operand), plus db =
// colsum(dz).  Replaces act_grad_colsum + a separate transpose_bf16 of dz
// (one full read of dz saved per layer per step, and one launch).
// 64x64 bf16 tiles staged in LDS at a 66-short (odd-dword) row stride:
// column reads in the transpose phase then hit 32 distinct banks.
__global__ __launch_bounds__(256)
void act_grad_colsum_T_kernel(const bf16* __restrict__ dy, const bf16* __restrict__ y,
                              bf16* __restrict__ dz, bf16* __restrict__ dzT,
                              float* __restrict__ db, long B, long N, int act) {
  __shared__ short tile[64 * 66];
  __shared__ float csum[64];
  long n0 = (long)blockIdx.x * 64, b0 = (long)blockIdx.y * 64;
  int t = threadIdx.x;
  if (t < 64) csum[t] = 0.0f;
  __syncthreads();

  const bool vec = (N % 8 == 0);
  int c8 = (t & 7) * 8;
  int r_ = t >> 3;
#pragma unroll
  for (int rr = 0; rr < 2; ++rr) {
    int r = r_ + rr * 32;
    long gb = b0 + r;
    s16x8 out;
    if (gb < B) {
      long base = gb * N + n0 + c8;
#pragma unroll
      for (int j = 0; j < 8; ++j) ((short*)&out)[j] = 0;
      if (vec && n0 + c8 + 8 <= N) {
        s16x8 vdy = *(const s16x8*)(dy + base);
        s16x8 vy = *(const s16x8*)(y + base);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = __bfloat162float(*(const bf16*)&((const short*)&vdy)[j]) *
                    act_grad_from_y(__bfloat162float(*(const bf16*)&((const short*)&vy)[j]), act);
          bf16 h = __float2bfloat16(g);
          ((short*)&out)[j] = *(short*)&h;
        }
        *(s16x8*)(dz + base) = out;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          long gn = n0 + c8 + j;
          if (gn < N) {
            float g = __bfloat162float(dy[gb * N + gn]) *
                      act_grad_from_y(__bfloat162float(y[gb * N + gn]), act);
            bf16 h = __float2bfloat16(g);
            ((short*)&out)[j] = *(short*)&h;
            dz[gb * N + gn] = h;
          }
        }
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) ((short*)&out)[j] = 0;
    }
#pragma unroll
    for (int j = 0; j < 8; j += 2)
      *(unsigned*)&tile[r * 66 + c8 + j] = *(unsigned*)&((short*)&out)[j];
  }
  __syncthreads();

  // transpose out + column sums: task = (column c, 8-row chunk)
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    int task = t + it * 256;
    int c = task >> 3;
    int r8 = (task & 7) * 8;
    s16x8 v;
    float part = 0.0f;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      short s = tile[(r8 + j) * 66 + c];
      ((short*)&v)[j] = s;
      part += __bfloat162float(*(bf16*)&s);
    }
    atomicAdd(&csum[c], part);
    long gn = n0 + c;
    long gb8 = b0 + r8;
    if (gn < N) {
      if (gb8 + 8 <= B) {
        *(s16x8*)(dzT + gn * B + gb8) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (gb8 + j < B) dzT[gn * B + gb8 + j] = *(bf16*)&((short*)&v)[j];
      }
    }
  }
  __syncthreads();
  if (t < 64 && n0 + t < N) atomicAdd(&db[n0 + t], csum[t]);
}

std::vector<at::Tensor> act_grad_colsum_T(at::Tensor dy, at::Tensor y, long act) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_BF16(dy);
  CHECK_GPU(y); CHECK_CONTIG(y); CHECK_BF16(y);
  long B = dy.size(0), N = dy.size(1);
  auto dz = at::empty_like(dy);
  auto dzT = at::empty({N, B}, dy.options());
  auto db = at::zeros({N}, dy.options().dtype(at::kFloat));
  dim3 grid((unsigned)((N + 63) / 64), (unsigned)((B + 63) / 64));
  hipLaunchKernelGGL(act_grad_colsum_T_kernel, grid, dim3(256), 0, cur_stream(),
                     (const bf16*)dy.data_ptr(), (const bf16*)y.data_ptr(),
                     (bf16*)dz.data_ptr(), (bf16*)dzT.data_ptr(),
                     (float*)db.data_ptr(), B, N, (int)act);
  return {dz, dzT, db};
}

// same, but the colsum atomically ACCUMULATES into db (the bias flat-grad
// view): no db zeros, no separate AccumulateGrad add
std::vector<at::Tensor> act_grad_colsum_T_into(at::Tensor dy, at::Tensor y,
                                               long act, at::Tensor db) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_BF16(dy);
  CHECK_GPU(y); CHECK_CONTIG(y); CHECK_BF16(y);
  CHECK_GPU(db); CHECK_F32(db);
  long B = dy.size(0), N = dy.size(1);
  TORCH_CHECK(db.numel() == N && db.is_contiguous(), "db view mismatch");
  auto dz = at::empty_like(dy);
  auto dzT = at::empty({N, B}, dy.options());
  dim3 grid((unsigned)((N + 63) / 64), (unsigned)((B + 63) / 64));
  hipLaunchKernelGGL(act_grad_colsum_T_kernel, grid, dim3(256), 0, cur_stream(),
                     (const bf16*)dy.data_ptr(), (const bf16*)y.data_ptr(),
                     (bf16*)dz.data_ptr(), (bf16*)dzT.data_ptr(),
                     (float*)db.data_ptr(), B, N, (int)act);
  return {dz, dzT};
}

// ---------------------------------------------------------------------------
// fused sigmoid + weighted loss (K3)
// fwd: p = sigmoid(z); per = w*(p-y)^2 (wmse) or w*bce (ce);
//      reduces loss_sum and wsum (block reduce + atomicAdd)
// bwd: dz = scale * w * dper/dz
// ---------------------------------------------------------------------------
enum LossKind { LOSS_WMSE = 0, LOSS_CE = 1 };

__global__ void loss_fwd_kernel(const bf16* __restrict__ z, const float* __restrict__ y,
                                const float* __restrict__ w, float* __restrict__ p,
                                float* __restrict__ loss_sum, float* __restrict__ wsum,
                                long n, int kind) {
  __shared__ float red[2][8];   // per-wave partials
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float ls = 0.0f, ws = 0.0f;
  for (long j = i; j < n; j += stride) {
    float zf = __bfloat162float(z[j]);
    float pf = 1.0f / (1.0f + __expf(-zf));
    p[j] = pf;
    float wf = w[j], yf = y[j];
    float per;
    if (kind == LOSS_WMSE) {
      float d = pf - yf;
      per = wf * d * d;
    } else {
      // stable bce-with-logits: max(z,0) - z*y + log1p(exp(-|z|))
      per = wf * (fmaxf(zf, 0.0f) - zf * yf + log1pf(__expf(-fabsf(zf))));
    }
    ls += per;
    // TF loss-reduction SUM_BY_NONZERO_WEIGHTS (the reference's
    // tf.losses.mean_squared_error default): normalize by the COUNT of
    // nonzero-weight samples, not sum(w)
    ws += (wf != 0.0f) ? 1.0f : 0.0f;
  }
  // wave reduce
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    ls += __shfl_down(ls, off, 64);
    ws += __shfl_down(ws, off, 64);
  }
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) { red[0][wave] = ls; red[1][wave] = ws; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float l = 0, s = 0;
    for (int wv = 0; wv < (int)(blockDim.x >> 6); ++wv) { l += red[0][wv]; s += red[1][wv]; }
    atomicAdd(loss_sum, l);
    atomicAdd(wsum, s);
  }
}

std::vector<at::Tensor> weighted_loss_fwd(at::Tensor z, at::Tensor y, at::Tensor w, long kind) {
  CHECK_GPU(z); CHECK_CONTIG(z); CHECK_BF16(z);
  CHECK_GPU(y); CHECK_CONTIG(y); CHECK_F32(y);
  CHECK_GPU(w); CHECK_CONTIG(w); CHECK_F32(w);
  long n = z.numel();
  auto p = at::empty({n}, z.options().dtype(at::kFloat));
  auto sums = at::zeros({2}, z.options().dtype(at::kFloat));
  auto loss_sum = sums.narrow(0, 0, 1).reshape({});
  auto wsum = sums.narrow(0, 1, 1).reshape({});
  int blocks = (int)std::min((n + 511) / 512 + 1, (long)1024);
  hipLaunchKernelGGL(loss_fwd_kernel, dim3(blocks), dim3(512), 0, cur_stream(),
                     (const bf16*)z.data_ptr(), (const float*)y.data_ptr(),
                     (const float*)w.data_ptr(), (float*)p.data_ptr(),
                     (float*)loss_sum.data_ptr(), (float*)wsum.data_ptr(), n, (int)kind);
  return {p, loss_sum, wsum};
}

__global__ void loss_bwd_kernel(const float* __restrict__ p, const float* __restrict__ y,
                                const float* __restrict__ w, bf16* __restrict__ dz,
                                long n, int kind, const float* __restrict__ scale_dev) {
  float scale = scale_dev[0];   // device scalar: no host sync, hipGraph-safe
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long j = i; j < n; j += stride) {
    float pf = p[j], yf = y[j], wf = w[j];
    float g = (kind == LOSS_WMSE) ? wf * 2.0f * (pf - yf) * pf * (1.0f - pf)
                                  : wf * (pf - yf);
    dz[j] = __float2bfloat16(g * scale);
  }
}

at::Tensor weighted_loss_bwd(at::Tensor p, at::Tensor y, at::Tensor w, long kind,
                             at::Tensor scale) {
  CHECK_GPU(p); CHECK_CONTIG(p); CHECK_F32(p);
  CHECK_F32(scale);
  long n = p.numel();
  auto dz = at::empty({n}, p.options().dtype(at::kBFloat16));
  int blocks = (int)std::min((n + 511) / 512 + 1, (long)1024);
  hipLaunchKernelGGL(loss_bwd_kernel, dim3(blocks), dim3(512), 0, cur_stream(),
                     (const float*)p.data_ptr(), (const float*)y.data_ptr(),
                     (const float*)w.data_ptr(), (bf16*)dz.data_ptr(),
                     n, (int)kind, (const float*)scale.data_ptr());
  return dz;
}

// ---------------------------------------------------------------------------
// K4 — fused optimizers over flat fp32 arena (g includes coupled L2 in-kernel)
// ---------------------------------------------------------------------------
__global__ void sgd_kernel(float* __restrict__ w, const float* __restrict__ g,
                           bf16* __restrict__ mir, long n, float lr, float l2) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = n / 4;
  for (long v = i; v < n4; v += stride) {
    f32x4 wv = ((f32x4*)w)[v], gv = ((const f32x4*)g)[v];
#pragma unroll
    for (int j = 0; j < 4; ++j) wv[j] -= lr * (gv[j] + l2 * wv[j]);
    ((f32x4*)w)[v] = wv;
    if (mir) {
      s16x4t pack;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        bf16 h = __float2bfloat16(wv[j]);
        ((short*)&pack)[j] = *(short*)&h;
      }
      *(s16x4t*)(mir + v * 4) = pack;
    }
  }
  for (long j = n4 * 4 + i; j < n; j += stride) {
    w[j] -= lr * (g[j] + l2 * w[j]);
    if (mir) mir[j] = __float2bfloat16(w[j]);
  }
}

__global__ void adam_kernel(float* __restrict__ w, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            bf16* __restrict__ mir,
                            long n, float lr, float b1, float b2, float eps,
                            float l2, float bc1, float sbc2) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float step = lr * sbc2 / bc1;
  for (long j = i; j < n; j += stride) {
    float gj = g[j] + l2 * w[j];
    float mj = b1 * m[j] + (1.0f - b1) * gj;
    float vj = b2 * v[j] + (1.0f - b2) * gj * gj;
    m[j] = mj; v[j] = vj;
    float wn = w[j] - step * mj / (sqrtf(vj) + eps * sbc2);
    w[j] = wn;
    if (mir) mir[j] = __float2bfloat16(wn);
  }
}

__global__ void adadelta_kernel(float* __restrict__ w, const float* __restrict__ g,
                                float* __restrict__ acc, float* __restrict__ dacc,
                                bf16* __restrict__ mir,
                                long n, float lr, float rho, float eps, float l2) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long j = i; j < n; j += stride) {
    float gj = g[j] + l2 * w[j];
    float a = rho * acc[j] + (1.0f - rho) * gj * gj;
    acc[j] = a;
    float upd = gj * sqrtf(dacc[j] + eps) / sqrtf(a + eps);
    dacc[j] = rho * dacc[j] + (1.0f - rho) * upd * upd;
    float wn = w[j] - lr * upd;
    w[j] = wn;
    if (mir) mir[j] = __float2bfloat16(wn);
  }
}

__global__ void adagrad_kernel(float* __restrict__ w, const float* __restrict__ g,
                               float* __restrict__ m, bf16* __restrict__ mir,
                               long n, float lr, float eps, float l2) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long j = i; j < n; j += stride) {
    float gj = g[j] + l2 * w[j];
    float mj = m[j] + gj * gj;
    m[j] = mj;
    float wn = w[j] - lr * gj / (sqrtf(mj) + eps);
    w[j] = wn;
    if (mir) mir[j] = __float2bfloat16(wn);
  }
}

static int opt_blocks(long n) { return (int)std::min((n + 1023) / 1024 + 1, (long)2048); }

static inline bf16* mir_ptr(at::Tensor& mir) {
  return mir.numel() ? (bf16*)mir.data_ptr() : nullptr;
}

void sgd_step(at::Tensor w, at::Tensor g, at::Tensor mir, double lr, double l2) {
  CHECK_GPU(w); CHECK_F32(w); CHECK_F32(g);
  hipLaunchKernelGGL(sgd_kernel, dim3(opt_blocks(w.numel())), dim3(256), 0, cur_stream(),
                     (float*)w.data_ptr(), (const float*)g.data_ptr(),
                     mir_ptr(mir), w.numel(), (float)lr, (float)l2);
}

void adam_step(at::Tensor w, at::Tensor g, at::Tensor m, at::Tensor v,
               at::Tensor mir,
               double lr, double b1, double b2, double eps, double l2, long t) {
  CHECK_GPU(w); CHECK_F32(w);
  float bc1 = 1.0f - powf((float)b1, (float)t);
  float sbc2 = sqrtf(1.0f - powf((float)b2, (float)t));
  hipLaunchKernelGGL(adam_kernel, dim3(opt_blocks(w.numel())), dim3(256), 0, cur_stream(),
                     (float*)w.data_ptr(), (const float*)g.data_ptr(),
                     (float*)m.data_ptr(), (float*)v.data_ptr(), mir_ptr(mir),
                     w.numel(),
                     (float)lr, (float)b1, (float)b2, (float)eps, (float)l2, bc1, sbc2);
}

// graph-safe Adam: the step counter lives on DEVICE so hipGraph replays see
// a fresh bias correction every replay (a host-computed bc would be frozen
// into the captured kernel args).
__global__ void step_incr_kernel(float* step_buf) {
  if (threadIdx.x == 0 && blockIdx.x == 0) step_buf[0] += 1.0f;
}

__global__ void adam_dev_kernel(float* __restrict__ w, const float* __restrict__ g,
                                float* __restrict__ m, float* __restrict__ v,
                                bf16* __restrict__ mir,
                                const float* __restrict__ step_buf, long n,
                                float lr, float b1, float b2, float eps, float l2) {
  float t = step_buf[0];
  float bc1 = 1.0f - powf(b1, t);
  float sbc2 = sqrtf(1.0f - powf(b2, t));
  float step = lr * sbc2 / bc1;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long j = i; j < n; j += stride) {
    float gj = g[j] + l2 * w[j];
    float mj = b1 * m[j] + (1.0f - b1) * gj;
    float vj = b2 * v[j] + (1.0f - b2) * gj * gj;
    m[j] = mj; v[j] = vj;
    float wn = w[j] - step * mj / (sqrtf(vj) + eps * sbc2);
    w[j] = wn;
    if (mir) mir[j] = __float2bfloat16(wn);
  }
}

void adam_step_dev(at::Tensor w, at::Tensor g, at::Tensor m, at::Tensor v,
                   at::Tensor mir, at::Tensor step_buf, double lr, double b1,
                   double b2, double eps, double l2) {
  CHECK_GPU(w); CHECK_F32(w); CHECK_F32(step_buf);
  auto s = cur_stream();
  hipLaunchKernelGGL(step_incr_kernel, dim3(1), dim3(64), 0, s,
                     (float*)step_buf.data_ptr());
  hipLaunchKernelGGL(adam_dev_kernel, dim3(opt_blocks(w.numel())), dim3(256), 0, s,
                     (float*)w.data_ptr(), (const float*)g.data_ptr(),
                     (float*)m.data_ptr(), (float*)v.data_ptr(), mir_ptr(mir),
                     (const float*)step_buf.data_ptr(), w.numel(),
                     (float)lr, (float)b1, (float)b2, (float)eps, (float)l2);
}

void adadelta_step(at::Tensor w, at::Tensor g, at::Tensor acc, at::Tensor dacc,
                   at::Tensor mir, double lr, double rho, double eps, double l2) {
  CHECK_GPU(w); CHECK_F32(w);
  hipLaunchKernelGGL(adadelta_kernel, dim3(opt_blocks(w.numel())), dim3(256), 0, cur_stream(),
                     (float*)w.data_ptr(), (const float*)g.data_ptr(),
                     (float*)acc.data_ptr(), (float*)dacc.data_ptr(), mir_ptr(mir),
                     w.numel(), (float)lr, (float)rho, (float)eps, (float)l2);
}

void adagrad_step(at::Tensor w, at::Tensor g, at::Tensor m, at::Tensor mir,
                  double lr, double eps, double l2) {
  CHECK_GPU(w); CHECK_F32(w);
  hipLaunchKernelGGL(adagrad_kernel, dim3(opt_blocks(w.numel())), dim3(256), 0, cur_stream(),
                     (float*)w.data_ptr(), (const float*)g.data_ptr(),
                     (float*)m.data_ptr(), mir_ptr(mir), w.numel(),
                     (float)lr, (float)eps, (float)l2);
}

// ---------------------------------------------------------------------------
// embedding arena gather: out[b, f*D+d] = arena[ids[b,f], d]  (bf16)
// one 16B (8 bf16) chunk per thread; out writes coalesced.
// ---------------------------------------------------------------------------
// out may be a STRIDED view: out[b] row starts at out + b*out_stride + col0
// (gather-into-concat: writes land directly inside the tower input buffer)
__global__ void emb_gather_kernel(const bf16* __restrict__ arena,
                                  const long* __restrict__ ids,
                                  bf16* __restrict__ out,
                                  long rows_bf, long D, long F,
                                  long out_stride, long col0) {
  long chunks_per_row = D / 8;
  long total = rows_bf * chunks_per_row;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = i; t < total; t += stride) {
    long rf = t / chunks_per_row;      // (b*F + f)
    long c = t % chunks_per_row;
    long row = ids[rf];
    long b = rf / F, f = rf % F;
    *(s16x8*)(out + b * out_stride + col0 + f * D + c * 8) =
        *(const s16x8*)(arena + row * D + c * 8);
  }
}

__global__ void emb_gather_scalar_kernel(const bf16* __restrict__ arena,
                                         const long* __restrict__ ids,
                                         bf16* __restrict__ out,
                                         long rows_bf, long D) {
  long total = rows_bf * D;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = i; t < total; t += stride) {
    long rf = t / D, d = t % D;
    out[t] = arena[ids[rf] * D + d];
  }
}

at::Tensor embedding_gather(at::Tensor arena, at::Tensor ids) {
  CHECK_GPU(arena); CHECK_CONTIG(arena); CHECK_BF16(arena);
  CHECK_GPU(ids); CHECK_CONTIG(ids);
  TORCH_CHECK(ids.scalar_type() == at::kLong, "ids must be int64");
  long B = ids.size(0), F = ids.size(1), D = arena.size(1);
  auto out = at::empty({B, F * D}, arena.options());
  long rows = B * F;
  if (D % 8 == 0) {
    long total = rows * (D / 8);
    int blocks = (int)std::min((total + 255) / 256 + 1, (long)4096);
    hipLaunchKernelGGL(emb_gather_kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                       (const bf16*)arena.data_ptr(), (const long*)ids.data_ptr(),
                       (bf16*)out.data_ptr(), rows, D, F, F * D, 0L);
  } else {
    long total = rows * D;
    int blocks = (int)std::min((total + 255) / 256 + 1, (long)4096);
    hipLaunchKernelGGL(emb_gather_scalar_kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                       (const bf16*)arena.data_ptr(), (const long*)ids.data_ptr(),
                       (bf16*)out.data_ptr(), rows, D);
  }
  return out;
}

// ---------------------------------------------------------------------------
// rowwise sparse embedding updates — SORT-FREE: rows may contain DUPLICATES
// (no torch.unique / rocprim sort on the hot path); correctness comes from
// atomics: global_atomic_pk_add_bf16 for the arena (unsafeAtomicAdd on
// __hip_bfloat162) and f32 atomicAdd for the adagrad accumulator.  Updates
// are linear in the gradient, so per-duplicate application == summed-grad
// application for a fixed denominator.
// ---------------------------------------------------------------------------
DEVINL void atomic_add_bf16_scalar(bf16* p, float v) {
  // 16-bit add via CAS on the containing aligned 32-bit word (odd-D tail)
  unsigned long addr = (unsigned long)p;
  unsigned* word = (unsigned*)(addr & ~3UL);
  int hi = (addr & 2) != 0;
  unsigned old = *word, assumed;
  do {
    assumed = old;
    unsigned short bits = hi ? (assumed >> 16) : (assumed & 0xffff);
    bf16 cur = *(bf16*)&bits;
    float nf = __bfloat162float(cur) + v;
    bf16 nb = __float2bfloat16(nf);
    unsigned short nbits = *(unsigned short*)&nb;
    unsigned repl = hi ? ((assumed & 0x0000ffffu) | ((unsigned)nbits << 16))
                       : ((assumed & 0xffff0000u) | nbits);
    old = atomicCAS(word, assumed, repl);
  } while (old != assumed);
}

// arena[rows[i]] += scale * vals[i] * rowscale[i] (rowscale nullable)
__global__ void emb_scatter_kernel(bf16* __restrict__ arena, const long* __restrict__ rows,
                                   const bf16* __restrict__ vals,
                                   const float* __restrict__ rowscale,
                                   long n, long D, float scale) {
  long pairs = D / 2;
  long total = n * (pairs ? pairs : 1);
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  if (pairs) {
    for (long t = i; t < total; t += stride) {
      long e = t / pairs, dp = t % pairs;
      float sc = scale * (rowscale ? rowscale[e] : 1.0f);
      long row = rows[e];
      __hip_bfloat162 add;
      add.x = __float2bfloat16(sc * __bfloat162float(vals[e * D + dp * 2]));
      add.y = __float2bfloat16(sc * __bfloat162float(vals[e * D + dp * 2 + 1]));
      unsafeAtomicAdd((__hip_bfloat162*)(arena + row * D + dp * 2), add);
    }
    if (D & 1) {  // odd tail element
      for (long e = i; e < n; e += stride) {
        float sc = scale * (rowscale ? rowscale[e] : 1.0f);
        atomic_add_bf16_scalar(arena + rows[e] * D + D - 1,
                               sc * __bfloat162float(vals[e * D + D - 1]));
      }
    }
  } else {  // D == 1
    for (long e = i; e < n; e += stride) {
      float sc = scale * (rowscale ? rowscale[e] : 1.0f);
      atomic_add_bf16_scalar(arena + rows[e], sc * __bfloat162float(vals[e]));
    }
  }
}

// generic scatter with the adagrad denominator inline: sc =
// -lr/(sqrt(acc[row])+eps) per entry (replaces emb_denom + rowscale buffer)
__global__ void emb_scatter_acc_kernel(bf16* __restrict__ arena, const long* __restrict__ rows,
                                       const bf16* __restrict__ vals,
                                       const float* __restrict__ acc,
                                       long n, long D, float scale, float eps) {
  long pairs = D / 2;
  long total = n * (pairs ? pairs : 1);
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  if (pairs) {
    for (long t = i; t < total; t += stride) {
      long e = t / pairs, dp = t % pairs;
      long row = rows[e];
      float sc = scale / (sqrtf(acc[row]) + eps);
      __hip_bfloat162 add;
      add.x = __float2bfloat16(sc * __bfloat162float(vals[e * D + dp * 2]));
      add.y = __float2bfloat16(sc * __bfloat162float(vals[e * D + dp * 2 + 1]));
      unsafeAtomicAdd((__hip_bfloat162*)(arena + row * D + dp * 2), add);
    }
    if (D & 1) {
      for (long e = i; e < n; e += stride) {
        long row = rows[e];
        float sc = scale / (sqrtf(acc[row]) + eps);
        atomic_add_bf16_scalar(arena + row * D + D - 1,
                               sc * __bfloat162float(vals[e * D + D - 1]));
      }
    }
  } else {
    for (long e = i; e < n; e += stride) {
      long row = rows[e];
      float sc = scale / (sqrtf(acc[row]) + eps);
      atomic_add_bf16_scalar(arena + row, sc * __bfloat162float(vals[e]));
    }
  }
}

// small-D phase 1: one THREAD per entry (a wave per entry would idle 63/64
// lanes at D=1 — the wide-column arena case)
__global__ void emb_accsq_small_kernel(float* __restrict__ acc, const long* __restrict__ rows,
                                       const bf16* __restrict__ vals, long n, long D) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long e = i; e < n; e += stride) {
    float sq = 0.0f;
    for (long d = 0; d < D; ++d) {
      float g = __bfloat162float(vals[e * D + d]);
      sq += g * g;
    }
    atomicAdd(&acc[rows[e]], sq / (float)D);
  }
}

// phase 1 of adagrad: acc[rows[i]] += mean_d vals[i,d]^2.
// A wave covers TWO entries (lanes 0-31 / 32-63), each lane loading a bf16
// PAIR (4 B) -> 256 B coalesced per wave access (a lane-per-element layout
// read only 2 B/lane: 0.56 TB/s measured; this is the fix).  Requires even D.
__global__ void emb_accsq_kernel(float* __restrict__ acc, const long* __restrict__ rows,
                                 const bf16* __restrict__ vals, long n, long D) {
  long e = (long)blockIdx.x * (blockDim.x >> 5) + (threadIdx.x >> 5);
  if (e >= n) return;
  int lane = threadIdx.x & 31;
  long pairs = D >> 1;
  float sq = 0.0f;
  const unsigned* v2 = (const unsigned*)(vals + e * D);
  for (long p = lane; p < pairs; p += 32) {
    unsigned u = v2[p];
    unsigned short lo = (unsigned short)(u & 0xffff), hi = (unsigned short)(u >> 16);
    float a = __bfloat162float(*(const bf16*)&lo);
    float b = __bfloat162float(*(const bf16*)&hi);
    sq += a * a + b * b;
  }
#pragma unroll
  for (int off = 16; off > 0; off >>= 1) sq += __shfl_down(sq, off, 32);
  if (lane == 0) atomicAdd(&acc[rows[e]], sq / (float)D);
}

// phase 2 denominator: rowscale[i] = 1/(sqrt(acc[rows[i]]) + eps)
__global__ void emb_denom_kernel(const float* __restrict__ acc, const long* __restrict__ rows,
                                 float* __restrict__ rowscale, long n, float eps) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long e = i; e < n; e += stride)
    rowscale[e] = 1.0f / (sqrtf(acc[rows[e]]) + eps);
}

static int scat_blocks(long total) {
  return (int)std::min((total + 255) / 256 + 1, (long)4096);
}

// gather arena rows for ids [B,F] directly into out[:, col0:col0+F*D]
// (out is the [B, n_dense + F*D] tower-input buffer; avoids a separate
// torch.cat pass over the concatenated activations)
void embedding_gather_into(at::Tensor arena, at::Tensor ids, at::Tensor out,
                           long col0) {
  CHECK_GPU(arena); CHECK_CONTIG(arena); CHECK_BF16(arena);
  CHECK_GPU(out); CHECK_CONTIG(out); CHECK_BF16(out);
  TORCH_CHECK(ids.scalar_type() == at::kLong, "ids must be int64");
  long B = ids.size(0), F = ids.size(1), D = arena.size(1);
  TORCH_CHECK(D % 8 == 0, "gather_into needs D%8==0");
  long out_stride = out.size(1);
  TORCH_CHECK(col0 + F * D <= out_stride, "slice out of range");
  long rows = B * F;
  long total = rows * (D / 8);
  int blocks = (int)std::min((total + 255) / 256 + 1, (long)4096);
  hipLaunchKernelGGL(emb_gather_kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                     (const bf16*)arena.data_ptr(), (const long*)ids.contiguous().data_ptr(),
                     (bf16*)out.data_ptr(), rows, D, F, out_stride, col0);
}

// Unified Wide&Deep arena gather (ROADMAP item 3): ONE [R, D+2] arena holds
// each category's deep D-vector (cols 0..D-1), its wide scalar weight
// (col D) and a zero pad col (D+1, keeps rows 4B-even for the pk-bf16
// atomics).  One pass writes the deep columns straight into the tower-input
// concat buffer and the wide column to a separate [B, F] buffer — replacing
// the round-1 pair of gathers over two arenas.
__global__ void emb_gather_split_kernel(const bf16* __restrict__ arena,
                                        const long* __restrict__ ids,
                                        bf16* __restrict__ out,
                                        bf16* __restrict__ wide,
                                        long rows_bf, long D, long DP, long F,
                                        long out_stride, long col0) {
  long chunks_per_row = D / 8;
  long deep_total = rows_bf * chunks_per_row;
  long total = deep_total + rows_bf;           // + one wide element per row
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  // 2-way unroll: the ids -> arena chain is a dependent random load (~82%
  // WAIT_ANY per PMC) and the capped grid makes each thread loop; issuing
  // two arena loads before either store doubles the random loads in flight.
  for (long t = i; t < total; t += 2 * stride) {
    long u = t + stride;
    s16x8 v1; bf16 w1; long o1 = -1, ow1 = 0;
    if (t < deep_total) {
      long rf = t / chunks_per_row;
      long c = t % chunks_per_row;
      long row = ids[rf];
      long b = rf / F, f = rf % F;
      v1 = *(const s16x8*)(arena + row * DP + c * 8);
      o1 = b * out_stride + col0 + f * D + c * 8;
    } else {
      long rf = t - deep_total;
      w1 = arena[ids[rf] * DP + D];
      ow1 = rf;
    }
    s16x8 v2; bf16 w2; long o2 = -1, ow2 = 0;
    if (u < total) {
      if (u < deep_total) {
        long rf = u / chunks_per_row;
        long c = u % chunks_per_row;
        long row = ids[rf];
        long b = rf / F, f = rf % F;
        v2 = *(const s16x8*)(arena + row * DP + c * 8);
        o2 = b * out_stride + col0 + f * D + c * 8;
      } else {
        long rf = u - deep_total;
        w2 = arena[ids[rf] * DP + D];
        ow2 = rf;
      }
    }
    if (o1 >= 0) *(s16x8*)(out + o1) = v1; else wide[ow1] = w1;
    if (u < total) {
      if (o2 >= 0) *(s16x8*)(out + o2) = v2; else wide[ow2] = w2;
    }
  }
}

// tower_in[:, col0:col0+F*D] and wide[B,F] from one [R, D+2] arena
void emb_gather_split(at::Tensor arena, at::Tensor ids, at::Tensor out,
                      at::Tensor wide, long col0, long D) {
  CHECK_GPU(arena); CHECK_CONTIG(arena); CHECK_BF16(arena);
  CHECK_GPU(out); CHECK_CONTIG(out); CHECK_BF16(out);
  CHECK_GPU(wide); CHECK_CONTIG(wide); CHECK_BF16(wide);
  TORCH_CHECK(ids.scalar_type() == at::kLong, "ids must be int64");
  long B = ids.size(0), F = ids.size(1), DP = arena.size(1);
  TORCH_CHECK(D % 8 == 0 && D < DP, "gather_split needs D%8==0 and D < arena cols");
  long out_stride = out.size(1);
  TORCH_CHECK(col0 + F * D <= out_stride, "slice out of range");
  TORCH_CHECK(wide.numel() == B * F, "wide buffer must be [B,F]");
  long rows = B * F;
  long total = rows * (D / 8) + rows;
  int blocks = (int)std::min((total + 255) / 256 + 1, (long)4096);
  hipLaunchKernelGGL(emb_gather_split_kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                     (const bf16*)arena.data_ptr(), (const long*)ids.contiguous().data_ptr(),
                     (bf16*)out.data_ptr(), (bf16*)wide.data_ptr(),
                     rows, D, DP, F, out_stride, col0);
}

// ---------------------------------------------------------------------------
// Unified-arena rowwise updates WITHOUT materializing a packed [n, D+2]
// gradient: the deep gradient is read straight from the tower-input grad
// buffer (entry e=(b,f): dgrad[b*dstride + dcol0 + f*D + d]) and the wide
// gradient from its own buffer with stride wstride.  The wide column updates
// through the SAME packed-bf16 atomic as the deep pairs — its pair partner
// is the zero pad column, fed a 0 addend.  Passing dgrad=packed_vals,
// dstride=F*(D+2), dcol0=0, wide=packed_vals+D, wstride=D+2 makes these the
// fast path for packed unified values too (the generic D=66 kernels lose
// >2x to the 132-byte row stride).
// ---------------------------------------------------------------------------
// accumulator addressing: addr = acc + row*astride + aoff.  External [R]
// array: astride=1, aoff=0.  IN-ARENA accumulator (f32 bits in bf16 cols
// D+2..D+3): acc = (float*)arena, astride=DP/2, aoff=(D+2)/2 — the atomic
// then lands in the same row the scatter updates.
__global__ void emb_accsq_uni_kernel(float* __restrict__ acc, const long* __restrict__ rows,
                                     const bf16* __restrict__ dgrad,
                                     const bf16* __restrict__ wide,
                                     long n, long F, long D, long DP,
                                     long dstride, long dcol0, long wstride,
                                     long astride, long aoff) {
  // two entries per 32-lane group with interleaved loads: doubles the
  // independent memory ops in flight (the kernel is ~82% WAIT_ANY)
  long g = (long)blockIdx.x * (blockDim.x >> 5) + (threadIdx.x >> 5);
  long e0 = g * 2, e1 = g * 2 + 1;
  if (e0 >= n) return;
  const bool two = (e1 < n);
  int lane = threadIdx.x & 31;
  long pairs = D >> 1;
  const unsigned* va = (const unsigned*)(dgrad + (e0 / F) * dstride + dcol0 + (e0 % F) * D);
  const unsigned* vb = two ? (const unsigned*)(dgrad + (e1 / F) * dstride + dcol0 + (e1 % F) * D) : va;
  float sq0 = 0.0f, sq1 = 0.0f;
  for (long p = lane; p < pairs; p += 32) {
    unsigned u0 = va[p];
    unsigned u1 = vb[p];
    unsigned short lo = (unsigned short)(u0 & 0xffff), hi = (unsigned short)(u0 >> 16);
    float a = __bfloat162float(*(const bf16*)&lo);
    float b = __bfloat162float(*(const bf16*)&hi);
    sq0 += a * a + b * b;
    lo = (unsigned short)(u1 & 0xffff); hi = (unsigned short)(u1 >> 16);
    a = __bfloat162float(*(const bf16*)&lo);
    b = __bfloat162float(*(const bf16*)&hi);
    sq1 += a * a + b * b;
  }
  if (lane == 0) {
    float w = __bfloat162float(wide[e0 * wstride]);
    sq0 += w * w;
    if (two) {
      float w2 = __bfloat162float(wide[e1 * wstride]);
      sq1 += w2 * w2;
    }
  }
#pragma unroll
  for (int off = 16; off > 0; off >>= 1) {
    sq0 += __shfl_down(sq0, off, 32);
    sq1 += __shfl_down(sq1, off, 32);
  }
  if (lane == 0) {
    atomicAdd(&acc[rows[e0] * astride + aoff], sq0 / (float)DP);
    if (two) atomicAdd(&acc[rows[e1] * astride + aoff], sq1 / (float)DP);
  }
}

__global__ void emb_scatter_uni_kernel(bf16* __restrict__ arena, const long* __restrict__ rows,
                                       const bf16* __restrict__ dgrad,
                                       const bf16* __restrict__ wide,
                                       const float* __restrict__ acc,
                                       long n, long F, long D, long DP,
                                       long dstride, long dcol0, long wstride,
                                       float scale, float eps,
                                       long astride, long aoff) {
  // 32 lanes per entry: lane k applies deep pairs k, k+32, ...; lane 0 also
  // the (wide, pad) pair.  Shift-only index math — a 64-bit `t / 33` per
  // element (the naive pairs-flattened mapping) measured ~2x slower.
  // The grid is capped (scat_blocks), so each group loops ~n/32768 entries;
  // the rows[e] -> acc[row] chain is two dependent RANDOM loads (~85%
  // WAIT_ANY per PMC), so the NEXT entry's chain is prefetched while the
  // current one is applied — legal to hoist past the arena atomics because
  // the scatter never touches the accumulator columns.
  long e = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 5;
  long estride = ((long)gridDim.x * blockDim.x) >> 5;
  int lane = threadIdx.x & 31;
  long hp = D >> 1;
  // 2-deep: row ids prefetched two iterations ahead, accumulators one —
  // each link of the rows -> acc chain then has a full iteration of slack,
  // so neither random-load latency sits on the critical path.
  long row = (e < n) ? rows[e] : 0;
  long row1 = (e + estride < n) ? rows[e + estride] : 0;
  float av = (acc && e < n) ? acc[row * astride + aoff] : 0.0f;
  for (; e < n; e += estride) {
    long e2 = e + 2 * estride;
    long row2 = (e2 < n) ? rows[e2] : 0;
    float av_n = (acc && e + estride < n) ? acc[row1 * astride + aoff] : 0.0f;
    // adagrad denominator inline (accsq pass completed): one broadcast
    // read replaces the separate emb_denom kernel + rowscale buffer
    float sc = acc ? scale / (sqrtf(av) + eps) : scale;
    const bf16* src = dgrad + (e / F) * dstride + dcol0 + (e % F) * D;
    bf16* dst = arena + row * DP;
    for (long dp = lane; dp < hp; dp += 32) {
      __hip_bfloat162 add;
      add.x = __float2bfloat16(sc * __bfloat162float(src[dp * 2]));
      add.y = __float2bfloat16(sc * __bfloat162float(src[dp * 2 + 1]));
      unsafeAtomicAdd((__hip_bfloat162*)(dst + dp * 2), add);
    }
    if (lane == 0) {
      __hip_bfloat162 add;
      add.x = __float2bfloat16(sc * __bfloat162float(wide[e * wstride]));
      add.y = __float2bfloat16(0.0f);    // pad column stays zero
      unsafeAtomicAdd((__hip_bfloat162*)(dst + D), add);
    }
    row = row1;
    row1 = row2;
    av = av_n;
  }
}

void emb_update_unified_binned(at::Tensor arena, at::Tensor acc, at::Tensor rows,
                               at::Tensor dgrad, long dcol0, at::Tensor wide,
                               long wstride, long F, double lr, double eps,
                               bool adagrad, bool acc_in_arena);

// SHIFU_EMB_BINNED=1 routes eligible unified updates through the atomic-free
// binned path (above); default off pending measurement.
static bool emb_binned_mode() {
  static int v = [] {
    const char* e = getenv("SHIFU_EMB_BINNED");
    return e ? atoi(e) : 0;
  }();
  return v != 0;
}

// adagrad (rowwise, fp32 accumulator) over a unified arena from unpacked
// grads; kind 0 = sgd (no accumulator)
void emb_update_unified(at::Tensor arena, at::Tensor acc, at::Tensor rows,
                        at::Tensor dgrad, long dcol0, at::Tensor wide,
                        long wstride, long F, double lr, double eps,
                        bool adagrad, bool acc_in_arena) {
  if (emb_binned_mode() && arena.size(1) <= 132 && arena.size(0) < (1L << 31)
      && rows.numel() < (1L << 31)) {
    emb_update_unified_binned(arena, acc, rows, dgrad, dcol0, wide, wstride,
                              F, lr, eps, adagrad, acc_in_arena);
    return;
  }
  CHECK_GPU(arena); CHECK_CONTIG(arena); CHECK_BF16(arena);
  CHECK_GPU(dgrad); CHECK_BF16(dgrad);
  CHECK_GPU(wide); CHECK_BF16(wide);
  long n = rows.numel();
  if (!n) return;
  long DP = arena.size(1);
  long D = DP - (acc_in_arena ? 4 : 2);
  long dstride = dgrad.size(1);
  TORCH_CHECK(D > 0 && D % 2 == 0, "unified update needs even deep D");
  TORCH_CHECK(dgrad.stride(1) == 1 && dgrad.stride(0) == dstride,
              "dgrad must be row-contiguous");
  TORCH_CHECK(dcol0 % 2 == 0 && dstride % 2 == 0,
              "unified update needs 4B-aligned deep columns");
  auto s = cur_stream();
  const float* accp = nullptr;
  float* accw = nullptr;
  long astride = 1, aoff = 0;
  if (acc_in_arena) {
    // D+4 layout: raw f32 accumulator in bf16 cols D+2..D+3 (4B-aligned
    // because D and DP are even)
    TORCH_CHECK(D % 2 == 0 && DP % 2 == 0, "acc_in_arena needs even D/DP");
    accw = (float*)arena.data_ptr();
    astride = DP / 2;
    aoff = (D + 2) / 2;
  } else if (adagrad) {
    CHECK_F32(acc);
    accw = (float*)acc.data_ptr();
  }
  if (adagrad) {
    int epb = 16;   // 8 groups x 2 entries per 256-thread block
    hipLaunchKernelGGL(emb_accsq_uni_kernel, dim3((unsigned)((n + epb - 1) / epb)),
                       dim3(256), 0, s,
                       accw, (const long*)rows.data_ptr(),
                       (const bf16*)dgrad.data_ptr(), (const bf16*)wide.data_ptr(),
                       n, F, D, DP, dstride, dcol0, wstride, astride, aoff);
    accp = accw;
  }
  long total = n * 32;   // 32 lanes per entry (shift-mapped)
  hipLaunchKernelGGL(emb_scatter_uni_kernel, dim3(scat_blocks(total)), dim3(256), 0, s,
                     (bf16*)arena.data_ptr(), (const long*)rows.data_ptr(),
                     (const bf16*)dgrad.data_ptr(), (const bf16*)wide.data_ptr(),
                     accp, n, F, D, DP, dstride, dcol0, wstride, (float)-lr,
                     (float)eps, astride, aoff);
}

// ---------------------------------------------------------------------------
// BINNED unified update — atomic-free owner-partitioned rewrite of
// accsq+scatter.  The atomic chain issues ~33 packed-bf16 atomics per entry
// (~28M per bench step), which bounds it on atomic issue/L2-RMW, not HBM.
// Here entries are bucketed by arena-row HIGH bits (kernels: count → scan →
// bin), so ONE workgroup owns every duplicate of a row; inside the workgroup
// same-row entries are chained through an LDS hash on the row LOW bits
// (unique within a bucket by construction — a bucket spans exactly the rows
// sharing its high bits).  One 32-lane group then applies a row's WHOLE
// update: f32-accumulate the chain's gradients (dgrad read ONCE — the
// separate accsq pass is gone), update the adagrad accumulator, and RMW the
// arena row with plain vector loads/stores.  Zero global atomics, and the
// denominator still sees the row's full-step accumulator because all of a
// row's entries sit in one chain.  (Only a bucket overflowing one 1024-entry
// chunk — pathological id skew — splits a chain; those rows then see
// partial-step denominators for the early chunks, which is still a valid
// adagrad-style step.)
// ---------------------------------------------------------------------------
#define EB_CH 1024
#define EB_HEAD 4096   // max bucket span (shift cap 12)

__global__ void emb_bin_count_kernel(const long* __restrict__ rows,
                                     int* __restrict__ counts, long n, int shift) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long e = i; e < n; e += stride)
    atomicAdd(&counts[rows[e] >> shift], 1);
}

// offsets[0..nb] = exclusive scan; cursor = copy of the starts (kernel C
// consumes it with atomics).  Single block, chunked Hillis-Steele.
__global__ __launch_bounds__(256)
void emb_bin_scan_kernel(const int* __restrict__ counts, int* __restrict__ offsets,
                         int* __restrict__ cursor, long nb) {
  __shared__ int buf[256];
  __shared__ int carry;
  int tid = threadIdx.x;
  if (tid == 0) { carry = 0; offsets[0] = 0; }
  __syncthreads();
  for (long c0 = 0; c0 < nb; c0 += 256) {
    int v = (c0 + tid < nb) ? counts[c0 + tid] : 0;
    buf[tid] = v;
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
      int add = (tid >= off) ? buf[tid - off] : 0;
      __syncthreads();
      buf[tid] += add;
      __syncthreads();
    }
    if (c0 + tid < nb) {
      int incl = carry + buf[tid];
      offsets[c0 + tid + 1] = incl;
      cursor[c0 + tid] = incl - v;
    }
    __syncthreads();
    if (tid == 0) carry += buf[255];
    __syncthreads();
  }
}

__global__ void emb_bin_scatter_kernel(const long* __restrict__ rows,
                                       int* __restrict__ cursor,
                                       unsigned* __restrict__ ebin,
                                       unsigned* __restrict__ rbin,
                                       long n, int shift) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long e = i; e < n; e += stride) {
    long r = rows[e];
    int slot = atomicAdd(&cursor[r >> shift], 1);
    ebin[slot] = (unsigned)e;
    rbin[slot] = (unsigned)r;
  }
}

// one block per bucket; 32-lane groups each own whole rows (D <= 128)
__global__ __launch_bounds__(256)
void emb_apply_binned_kernel(bf16* __restrict__ arena,
                             const unsigned* __restrict__ ebin,
                             const unsigned* __restrict__ rbin,
                             const int* __restrict__ offsets,
                             const bf16* __restrict__ dgrad,
                             const bf16* __restrict__ wide,
                             float* __restrict__ accw,
                             long F, long D, long DP,
                             long dstride, long dcol0, long wstride,
                             float scale, float eps, int shift,
                             long astride, long aoff) {
  __shared__ unsigned s_e[EB_CH];
  __shared__ unsigned s_row[EB_CH];
  __shared__ int s_next[EB_CH];
  __shared__ int s_head[EB_HEAD];
  __shared__ int s_work[EB_CH];
  __shared__ int s_nwork;
  const int b = blockIdx.x;
  const int start = offsets[b], end = offsets[b + 1];
  if (start >= end) return;
  const int tid = threadIdx.x;
  const int mask = (1 << shift) - 1;
  const int warp = tid >> 5, lane = tid & 31;
  const long pairs = D >> 1;

  for (int c0 = start; c0 < end; c0 += EB_CH) {
    int cnt = min(EB_CH, end - c0);
    for (int i = tid; i < EB_HEAD; i += 256) s_head[i] = -1;
    if (tid == 0) s_nwork = 0;
    __syncthreads();
    for (int j = tid; j < cnt; j += 256) {
      unsigned e = ebin[c0 + j];
      unsigned r = rbin[c0 + j];
      s_e[j] = e;
      s_row[j] = r;
      int old = atomicExch(&s_head[r & mask], j);
      s_next[j] = old;
      if (old < 0) {                       // first occurrence claims the row
        int w = atomicAdd(&s_nwork, 1);
        s_work[w] = (int)(r & mask);
      }
    }
    __syncthreads();
    const int nw = s_nwork;
    for (int w = warp; w < nw; w += 8) {
      const int hl = s_work[w];
      float ga[2][2] = {{0.f, 0.f}, {0.f, 0.f}};
      float sqsum = 0.f, gw = 0.f;
      for (int j = s_head[hl]; j >= 0; j = s_next[j]) {
        unsigned e = s_e[j];
        const unsigned* v2 = (const unsigned*)(dgrad + (long)(e / F) * dstride +
                                               dcol0 + (long)(e % F) * D);
#pragma unroll
        for (int pi = 0; pi < 2; ++pi) {
          long p = lane + pi * 32;
          if (p < pairs) {
            unsigned u = v2[p];
            unsigned short lo = (unsigned short)(u & 0xffff);
            unsigned short hi = (unsigned short)(u >> 16);
            float a = __bfloat162float(*(const bf16*)&lo);
            float c = __bfloat162float(*(const bf16*)&hi);
            ga[pi][0] += a;
            ga[pi][1] += c;
            sqsum += a * a + c * c;
          }
        }
        if (lane == 0) {
          float wv = __bfloat162float(wide[(long)e * wstride]);
          gw += wv;
          sqsum += wv * wv;
        }
      }
      const long row = (long)s_row[s_head[hl]];
      float sc = scale;
      if (accw) {
#pragma unroll
        for (int off = 16; off > 0; off >>= 1) sqsum += __shfl_down(sqsum, off, 32);
        if (lane == 0) {
          float* ap = &accw[row * astride + aoff];
          float anew = *ap + sqsum / (float)DP;
          *ap = anew;                       // exclusive owner: plain RMW
          sc = scale / (sqrtf(anew) + eps);
        }
        sc = __shfl(sc, 0, 32);
      }
      unsigned* d32 = (unsigned*)(arena + row * DP);
#pragma unroll
      for (int pi = 0; pi < 2; ++pi) {
        long p = lane + pi * 32;
        if (p < pairs) {
          unsigned u = d32[p];
          unsigned short lo = (unsigned short)(u & 0xffff);
          unsigned short hi = (unsigned short)(u >> 16);
          bf16 nl = __float2bfloat16(__bfloat162float(*(const bf16*)&lo) + sc * ga[pi][0]);
          bf16 nh = __float2bfloat16(__bfloat162float(*(const bf16*)&hi) + sc * ga[pi][1]);
          d32[p] = (unsigned)*(unsigned short*)&nl | ((unsigned)*(unsigned short*)&nh << 16);
        }
      }
      if (lane == 0) {                      // (wide, pad) pair — pad untouched
        unsigned u = d32[pairs];
        unsigned short lo = (unsigned short)(u & 0xffff);
        bf16 nl = __float2bfloat16(__bfloat162float(*(const bf16*)&lo) + sc * gw);
        d32[pairs] = (u & 0xffff0000u) | (unsigned)*(unsigned short*)&nl;
      }
    }
    __syncthreads();
  }
}

// same contract as emb_update_unified, atomic-free binned implementation
void emb_update_unified_binned(at::Tensor arena, at::Tensor acc, at::Tensor rows,
                               at::Tensor dgrad, long dcol0, at::Tensor wide,
                               long wstride, long F, double lr, double eps,
                               bool adagrad, bool acc_in_arena) {
  CHECK_GPU(arena); CHECK_CONTIG(arena); CHECK_BF16(arena);
  CHECK_GPU(dgrad); CHECK_BF16(dgrad);
  CHECK_GPU(wide); CHECK_BF16(wide);
  long n = rows.numel();
  if (!n) return;
  long R = arena.size(0), DP = arena.size(1);
  long D = DP - (acc_in_arena ? 4 : 2);
  long dstride = dgrad.size(1);
  TORCH_CHECK(D > 0 && D % 2 == 0 && D <= 128, "binned update needs even D <= 128");
  TORCH_CHECK(R < (1L << 31) && n < (1L << 31), "binned update: 32-bit ids");
  TORCH_CHECK(dgrad.stride(1) == 1 && dgrad.stride(0) == dstride,
              "dgrad must be row-contiguous");
  TORCH_CHECK(dcol0 % 2 == 0 && dstride % 2 == 0,
              "unified update needs 4B-aligned deep columns");
  auto s = cur_stream();
  float* accw = nullptr;
  long astride = 1, aoff = 0;
  if (acc_in_arena) {
    accw = (float*)arena.data_ptr();
    astride = DP / 2;
    aoff = (D + 2) / 2;
  } else if (adagrad) {
    CHECK_F32(acc);
    accw = (float*)acc.data_ptr();
  }
  if (!adagrad) accw = nullptr;

  // bucket span: largest power of two <= R/(n/256) so buckets average ~256
  // entries (one LDS chunk), capped at the head-table size
  long nb_t = std::max(n / 256, 1L);
  long span = std::max(R / nb_t, 1L);
  int shift = 0;
  while ((2L << shift) <= span && shift < 12) shift++;
  long nb = (R >> shift) + 1;

  auto iopt = arena.options().dtype(at::kInt);
  auto counts = at::zeros({nb}, iopt);
  auto offsets = at::empty({nb + 1}, iopt);
  auto cursor = at::empty({nb}, iopt);
  auto ebin = at::empty({n}, iopt);
  auto rbin = at::empty({n}, iopt);
  int blocks = (int)std::min((n + 255) / 256, 4096L);
  const long* rp = (const long*)rows.data_ptr();
  hipLaunchKernelGGL(emb_bin_count_kernel, dim3(blocks), dim3(256), 0, s,
                     rp, (int*)counts.data_ptr(), n, shift);
  hipLaunchKernelGGL(emb_bin_scan_kernel, dim3(1), dim3(256), 0, s,
                     (const int*)counts.data_ptr(), (int*)offsets.data_ptr(),
                     (int*)cursor.data_ptr(), nb);
  hipLaunchKernelGGL(emb_bin_scatter_kernel, dim3(blocks), dim3(256), 0, s,
                     rp, (int*)cursor.data_ptr(),
                     (unsigned*)ebin.data_ptr(), (unsigned*)rbin.data_ptr(),
                     n, shift);
  hipLaunchKernelGGL(emb_apply_binned_kernel, dim3((unsigned)nb), dim3(256), 0, s,
                     (bf16*)arena.data_ptr(),
                     (const unsigned*)ebin.data_ptr(), (const unsigned*)rbin.data_ptr(),
                     (const int*)offsets.data_ptr(),
                     (const bf16*)dgrad.data_ptr(), (const bf16*)wide.data_ptr(),
                     accw, F, D, DP, dstride, dcol0, wstride,
                     (float)-lr, (float)eps, shift, astride, aoff);
}

void emb_sgd_step(at::Tensor arena, at::Tensor rows, at::Tensor vals, double lr) {
  CHECK_GPU(arena); CHECK_BF16(arena); CHECK_BF16(vals);
  long n = rows.numel(), D = arena.size(1);
  if (!n) return;
  long total = n * std::max<long>(D / 2, 1);
  hipLaunchKernelGGL(emb_scatter_kernel, dim3(scat_blocks(total)), dim3(256), 0, cur_stream(),
                     (bf16*)arena.data_ptr(), (const long*)rows.data_ptr(),
                     (const bf16*)vals.data_ptr(), nullptr, n, D, (float)-lr);
}

void emb_adagrad_step(at::Tensor arena, at::Tensor acc, at::Tensor rows, at::Tensor vals,
                      double lr, double eps) {
  CHECK_GPU(arena); CHECK_BF16(arena); CHECK_F32(acc); CHECK_BF16(vals);
  long n = rows.numel(), D = arena.size(1);
  if (!n) return;
  auto s = cur_stream();
  if (D <= 8 || (D & 1)) {
    hipLaunchKernelGGL(emb_accsq_small_kernel, dim3(scat_blocks(n)), dim3(256), 0, s,
                       (float*)acc.data_ptr(), (const long*)rows.data_ptr(),
                       (const bf16*)vals.data_ptr(), n, D);
  } else {
    int epb = 8;  // 256 threads = 8 waves-of-32 lanes... 8 half-waves/block
    hipLaunchKernelGGL(emb_accsq_kernel, dim3((unsigned)((n + epb - 1) / epb)),
                       dim3(256), 0, s,
                       (float*)acc.data_ptr(), (const long*)rows.data_ptr(),
                       (const bf16*)vals.data_ptr(), n, D);
  }
  long total = n * std::max<long>(D / 2, 1);
  hipLaunchKernelGGL(emb_scatter_acc_kernel, dim3(scat_blocks(total)), dim3(256), 0, s,
                     (bf16*)arena.data_ptr(), (const long*)rows.data_ptr(),
                     (const bf16*)vals.data_ptr(), (const float*)acc.data_ptr(),
                     n, D, (float)-lr, (float)eps);
}

// ---------------------------------------------------------------------------
// GEMV path for 1-unit heads (shifu_output_0 / wide linear): a 128x128 MFMA
// tile on N=1 wastes 127/128 of the math — these run at memory speed.
// ---------------------------------------------------------------------------
// y[b] = act(sum_k x[b,k]*w[k] + bias) : one wave per row, shuffle-reduce
__global__ __launch_bounds__(256)
void gemv_fwd_kernel(const bf16* __restrict__ x, const bf16* __restrict__ w,
                     const bf16* __restrict__ bias, bf16* __restrict__ y,
                     long B, long K, int act) {
  long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= B) return;
  int lane = threadIdx.x & 63;
  float acc = 0.0f;
  for (long k = lane; k < K; k += 64)
    acc += __bfloat162float(x[row * K + k]) * __bfloat162float(w[k]);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  if (lane == 0) y[row] = __float2bfloat16(act_fwd(acc + __bfloat162float(bias[0]), act));
}

// dw[k] = sum_b x[b,k]*dz[b] ; db = sum_b dz[b]  (column-parallel + atomics)
// slab reduce for SMALL outputs with MANY z-parts (gemv dw|db: MN ~ a few
// hundred, z up to 2048): 64 lanes per output element, parallel over z —
// splitk_reduce's one-lane-per-element loop would serialize 2048 loads.
__global__ void slab_reduce_wide_kernel(const float* __restrict__ W,
                                        float* __restrict__ C, long MN, int z) {
  long e = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (e >= MN) return;
  int lane = threadIdx.x & 63;
  float s = 0.0f;
  for (int zz = lane; zz < z; zz += 64) s += W[(long)zz * MN + e];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off, 64);
  if (lane == 0) C[e] = s;
}

// slab variant: thread t owns 8 columns (16B vector loads — the old 2B
// per-thread column walk issued 8x the load instructions and measured ~10x
// off roofline); each row-chunk block plain-stores its [K+1] partial
// (dw | db) into wslab and splitk_reduce combines — no f32 atomics, bitwise
// deterministic (the atomic version's dw contention is what made the finer
// row chop regress in round 1).
__global__ void gemv_wgrad_kernel(const bf16* __restrict__ x, const bf16* __restrict__ dz,
                                  float* __restrict__ wslab,
                                  long B, long K, long rows_per_chunk) {
  long b0 = (long)blockIdx.x * rows_per_chunk;
  long b1 = min(B, b0 + rows_per_chunk);
  float* out = wslab + (long)blockIdx.x * (K + 1);
  long kc = (K + 7) >> 3;
  if ((K & 7) == 0) {
    for (long t = threadIdx.x; t < kc; t += blockDim.x) {
      long k8 = t * 8;
      float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
      for (long b = b0; b < b1; ++b) {
        float d = __bfloat162float(dz[b]);
        s16x8 v = *(const s16x8*)(x + b * K + k8);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[j] += d * __bfloat162float(*(const bf16*)&((const short*)&v)[j]);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) out[k8 + j] = acc[j];
    }
  } else {
    // odd K: per-column mapping (one scalar load per lane per row)
    for (long k = threadIdx.x; k < K; k += blockDim.x) {
      float acc = 0.0f;
      for (long b = b0; b < b1; ++b)
        acc += __bfloat162float(dz[b]) * __bfloat162float(x[b * K + k]);
      out[k] = acc;
    }
  }
  if (threadIdx.x < 64) {   // one wave reduces this chunk's db partial
    float acc = 0.0f;
    for (long b = b0 + threadIdx.x; b < b1; b += 64)
      acc += __bfloat162float(dz[b]);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if (threadIdx.x == 0) out[K] = acc;
  }
}

// dx[b,k] = dz[b] * w[k]  (outer product, vectorized by 8)
__global__ void gemv_dgrad_kernel(const bf16* __restrict__ dz, const bf16* __restrict__ w,
                                  bf16* __restrict__ dx, long B, long K) {
  long chunks = K / 8;
  long total = B * chunks;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = i; t < total; t += stride) {
    long b = t / chunks, c = (t % chunks) * 8;
    float d = __bfloat162float(dz[b]);
    s16x8 wv = *(const s16x8*)(w + c);
    s16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      ((bf16*)&out)[j] = __float2bfloat16(d * __bfloat162float(((const bf16*)&wv)[j]));
    *(s16x8*)(dx + b * K + c) = out;
  }
  // scalar tail columns
  for (long b = i; b < B && (K & 7); b += stride) {
    float d = __bfloat162float(dz[b]);
    for (long k = chunks * 8; k < K; ++k)
      dx[b * K + k] = __float2bfloat16(d * __bfloat162float(w[k]));
  }
}

at::Tensor gemv_fwd(at::Tensor x, at::Tensor w, at::Tensor bias, long act) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  long B = x.size(0), K = x.size(1);
  auto y = at::empty({B, 1}, x.options());
  int wpb = 4;
  hipLaunchKernelGGL(gemv_fwd_kernel, dim3((unsigned)((B + wpb - 1) / wpb)),
                     dim3(64 * wpb), 0, cur_stream(),
                     (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
                     (const bf16*)bias.data_ptr(), (bf16*)y.data_ptr(), B, K, (int)act);
  return y;
}

std::vector<at::Tensor> gemv_bwd(at::Tensor x, at::Tensor w, at::Tensor dz,
                                 bool need_dx) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
  long B = x.size(0), K = x.size(1);
  // dw|db in one buffer; per-chunk partials go to a [chunks, K+1] slab and
  // splitk_reduce combines (no memset, no atomics, deterministic)
  auto wb = at::empty({K + 1}, x.options().dtype(at::kFloat));
  auto dw = wb.narrow(0, 0, K).view({1, K});
  auto db = wb.narrow(0, K, 1);
  // 64-thread blocks (one wave) x 2048 chunks: same resident-wave count as
  // the old 256x512 grid — the vector loads alone regressed 4x when packed
  // into 1/4 the waves (memory parallelism, not instruction count, rules)
  long chunks = std::min<long>(2048, std::max<long>(B / 16, 1));
  long rpc = (B + chunks - 1) / chunks;
  auto wslab = at::empty({chunks, K + 1}, x.options().dtype(at::kFloat));
  hipLaunchKernelGGL(gemv_wgrad_kernel, dim3((unsigned)chunks), dim3(64), 0,
                     cur_stream(),
                     (const bf16*)x.data_ptr(), (const bf16*)dz.data_ptr(),
                     (float*)wslab.data_ptr(), B, K, rpc);
  hipLaunchKernelGGL(slab_reduce_wide_kernel,
                     dim3((unsigned)((K + 1 + 3) / 4)), dim3(256), 0,
                     cur_stream(), (const float*)wslab.data_ptr(),
                     (float*)wb.data_ptr(), K + 1, (int)chunks);
  at::Tensor dx;
  if (need_dx) {
    dx = at::empty({B, K}, x.options());
    long total = B * std::max<long>(K / 8, 1);
    hipLaunchKernelGGL(gemv_dgrad_kernel, dim3(scat_blocks(total)), dim3(256),
                       0, cur_stream(),
                       (const bf16*)dz.data_ptr(), (const bf16*)w.data_ptr(),
                       (bf16*)dx.data_ptr(), B, K);
  } else {
    dx = dw;  // placeholder; caller ignores dx when need_dx=false
  }
  return {dw, db, dx};
}

// ---------------------------------------------------------------------------
// DeepFM second-order interaction (fused):
//   fm2[b] = 0.5 * sum_d [ (sum_f v[b,f,d])^2 - sum_f v[b,f,d]^2 ]
// fwd: one wave per row b, lane d accumulates over F; saves s[b,d]=sum_f v
// bwd: dv[b,f,d] = (s[b,d] - v[b,f,d]) * dout[b]
// ---------------------------------------------------------------------------
__global__ void fm2_fwd_kernel(const bf16* __restrict__ emb, float* __restrict__ fm2,
                               float* __restrict__ s, long B, long F, long D,
                               long row_stride) {
  long b = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (b >= B) return;
  int lane = threadIdx.x & 63;
  float acc = 0.0f;
  for (long d = lane; d < D; d += 64) {
    float sum = 0.0f, sq = 0.0f;
    for (long f = 0; f < F; ++f) {
      float v = __bfloat162float(emb[b * row_stride + f * D + d]);
      sum += v;
      sq += v * v;
    }
    s[b * D + d] = sum;
    acc += sum * sum - sq;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  if (lane == 0) fm2[b] = 0.5f * acc;
}

__global__ void fm2_bwd_kernel(const bf16* __restrict__ emb, const float* __restrict__ s,
                               const float* __restrict__ dout, bf16* __restrict__ demb,
                               long B, long F, long D, long row_stride) {
  long total = B * F * D;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = i; t < total; t += stride) {
    long b = t / (F * D);
    long r = t % (F * D);
    long d = r % D;
    float v = __bfloat162float(emb[b * row_stride + r]);
    demb[t] = __float2bfloat16((s[b * D + d] - v) * dout[b]);
  }
}

std::vector<at::Tensor> fm2_fwd(at::Tensor emb, long F, long D) {
  // emb: [B, F*D] contiguous, or a row-strided view [B, F*D] of a wider
  // contiguous buffer (the fused tower-input concat)
  CHECK_GPU(emb); CHECK_BF16(emb);
  TORCH_CHECK(emb.stride(1) == 1, "inner dim must be contiguous");
  long B = emb.size(0), rs = emb.stride(0);
  auto fm2 = at::empty({B}, emb.options().dtype(at::kFloat));
  auto sum = at::empty({B, D}, emb.options().dtype(at::kFloat));
  int wpb = 4;
  hipLaunchKernelGGL(fm2_fwd_kernel, dim3((unsigned)((B + wpb - 1) / wpb)),
                     dim3(64 * wpb), 0, cur_stream(),
                     (const bf16*)emb.data_ptr(), (float*)fm2.data_ptr(),
                     (float*)sum.data_ptr(), B, F, D, rs);
  return {fm2, sum};
}

at::Tensor fm2_bwd(at::Tensor emb, at::Tensor s, at::Tensor dout, long F, long D) {
  CHECK_GPU(emb); CHECK_BF16(emb);
  TORCH_CHECK(emb.stride(1) == 1, "inner dim must be contiguous");
  CHECK_F32(s); CHECK_F32(dout);
  long B = emb.size(0), rs = emb.stride(0);
  auto demb = at::empty({B, F * D}, emb.options());
  long total = B * F * D;
  hipLaunchKernelGGL(fm2_bwd_kernel, dim3(scat_blocks(total)), dim3(256), 0, cur_stream(),
                     (const bf16*)emb.data_ptr(), (const float*)s.data_ptr(),
                     (const float*)dout.contiguous().data_ptr(),
                     (bf16*)demb.data_ptr(), B, F, D, rs);
  return demb;
}

// ---------------------------------------------------------------------------
PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "shifu_amd CDNA4 (gfx950) kernels";
  m.def("linear_act_fwd", &linear_act_fwd, "fused GEMM+bias+act forward (bf16)");
  m.def("gemm_nn_bf16", &gemm_nn_bf16);
  m.def("gemm_nt_bf16", &gemm_nt_bf16);
  m.def("gemm_tn_f32", &gemm_tn_f32);
  m.def("linear_nt_fwd", &linear_nt_fwd);
  m.def("gemm_ntv3_bf16", &gemm_ntv3_bf16);
  m.def("gemm_ntv3_f32", &gemm_ntv3_f32);
  m.def("gemm_ntv3_f32_into", &gemm_ntv3_f32_into);
  m.def("transpose_bf16", &transpose_bf16);
  m.def("gemm_tt_f32", &gemm_tt_f32);
  m.def("gemm_ttv3_f32", &gemm_ttv3_f32);
  m.def("gemm_ttv3_f32_into", &gemm_ttv3_f32_into);
  m.def("gemv_fwd", &gemv_fwd);
  m.def("gemv_bwd", &gemv_bwd);
  m.def("fm2_fwd", &fm2_fwd);
  m.def("fm2_bwd", &fm2_bwd);
  m.def("mfma_probe", &mfma_probe);
  m.def("mfma_probe32", &mfma_probe32);
  m.def("act_grad", &act_grad);
  m.def("colsum_f32", &colsum_f32);
  m.def("act_grad_colsum", &act_grad_colsum);
  m.def("act_grad_colsum_into", &act_grad_colsum_into);
  m.def("act_grad_colsum_T", &act_grad_colsum_T);
  m.def("act_grad_colsum_T_into", &act_grad_colsum_T_into);
  m.def("weighted_loss_fwd", &weighted_loss_fwd);
  m.def("weighted_loss_bwd", &weighted_loss_bwd);
  m.def("sgd_step", &sgd_step);
  m.def("adam_step", &adam_step);
  m.def("adam_step_dev", &adam_step_dev);
  m.def("adadelta_step", &adadelta_step);
  m.def("adagrad_step", &adagrad_step);
  m.def("embedding_gather", &embedding_gather);
  m.def("embedding_gather_into", &embedding_gather_into);
  m.def("emb_gather_split", &emb_gather_split);
  m.def("emb_update_unified", &emb_update_unified);
  m.def("emb_update_unified_binned", &emb_update_unified_binned);
  m.def("emb_sgd_step", &emb_sgd_step);
  m.def("emb_adagrad_step", &emb_adagrad_step);
}
