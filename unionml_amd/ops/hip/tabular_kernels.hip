// CDNA4 (gfx950 / MI355X) kernels for the default tabular hot path.
//
// Implements the kernel set of SURVEY.md §2c (net-new designs; the
// reference unionai-oss/unionml is pure Python and has no kernels):
//   - standardize_fit / standardize_apply : per-column (x-mean)*invstd, fp32 -> bf16
//   - mlp_step        : fused fwd+bwd of the digits MLP
//                       (IN=64 -> HID=32 relu -> CLS=10 softmax/xent)
//                       producing fp32 grads + loss in ONE launch (DP path:
//                       an RCCL all-reduce of the 2.6k-float grad buffer sits
//                       between this and adam_step)
//   - adam_step       : single-block fused Adam on the flat fp32 master,
//                       emitting the bf16 compute mirror
//   - mlp_train_steps : the single-GPU flagship — a persistent single-
//                       workgroup kernel running N optimizer steps in ONE
//                       launch: bf16 weights, fp32 master and Adam moments
//                       all live in LDS across steps; only the minibatch
//                       rows stream from HBM. Removes every per-step launch,
//                       zeroing pass, and HBM weight round-trip.
//   - mlp_predict     : fused standardize + fwd + argmax (serving hot path,
//                       hipGraph-captured per batch bucket)
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//   * wave64; training blocks = 8 waves (512 threads); each wave owns 16
//     batch rows per 128-row chunk.
//   * GEMM-shaped work on MFMA: __builtin_amdgcn_mfma_f32_16x16x32_bf16
//     (gfx950 2xK form), fp32 accumulate. Fragment mapping (verified on
//     hardware by tests/test_gpu_kernels.py with asymmetric operands):
//       A[16x32]:  lane l holds A[l&15][(l>>4)*8 + i], i = 0..7
//       B[32x16]:  lane l holds B[(l>>4)*8 + i][l&15]
//       C/D[16x16]: lane l, reg r holds D[(l>>4)*4 + r][l&15]
//   * LDS rows padded to strides whose ds_read_b128 bank slots are all
//     distinct (stride 144 B for X, 80 B for H/dL/dH) and 16 B aligned
//     (guide §6 G4/G17).
//   * Every launch is stream-ordered and hipGraph-capturable (no mallocs,
//     no syncs — guide Guideline 9). Step counters live in device memory
//     so Adam bias correction stays exact under graph replay.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define IN 64
#define HID 32
#define CLS 10
#define CPAD 16          // CLS padded to one MFMA tile
#define ROWS 128         // batch rows per workgroup / per chunk
#define WAVES 8
#define BLOCK (WAVES * 64)

// LDS row strides (in bf16 elements). Both keep every b128 A-fragment
// address 16 B aligned and give 16 distinct bank slots over 16 rows.
#define XS 72            // IN + 8   -> 144 B rows
#define HS 40            // HID + 8  -> 80 B rows
#define W2S 24           // CPAD + 8 (scalar reads only)

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef unsigned short u16;

__device__ __forceinline__ float bf2f(u16 v) {
  union { float f; unsigned u; } c;
  c.u = ((unsigned)v) << 16;
  return c.f;
}

__device__ __forceinline__ u16 f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;        // round-to-nearest-even
  return (u16)(c.u >> 16);
}

// ---------------------------------------------------------------------------
// standardize
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
standardize_fit_kernel(const float* __restrict__ X, long long N, int D,
                       float* __restrict__ mean, float* __restrict__ invstd,
                       float eps) {
  const int col = blockIdx.x;            // one workgroup per column
  if (col >= D) return;
  double s = 0.0, s2 = 0.0;
  for (long long r = threadIdx.x; r < N; r += blockDim.x) {
    const double v = (double)X[r * D + col];
    s += v;
    s2 += v * v;
  }
  __shared__ double ls[256], ls2[256];
  ls[threadIdx.x] = s;
  ls2[threadIdx.x] = s2;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      ls[threadIdx.x] += ls[threadIdx.x + off];
      ls2[threadIdx.x] += ls2[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    const double m = ls[0] / (double)N;
    const double var = ls2[0] / (double)N - m * m;
    mean[col] = (float)m;
    invstd[col] = (float)(1.0 / sqrt(var > 0.0 ? var + (double)eps : (double)eps));
  }
}

extern "C" __global__ void __launch_bounds__(256)
standardize_apply_kernel(const float* __restrict__ X, long long n_elems, int D,
                         const float* __restrict__ mean,
                         const float* __restrict__ invstd,
                         u16* __restrict__ out) {
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n_elems;
       i += (long long)gridDim.x * blockDim.x) {
    const int col = (int)(i % D);
    out[i] = f2bf((X[i] - mean[col]) * invstd[col]);
  }
}

// ---------------------------------------------------------------------------
// flat master/grads layout (floats):
//   [0, 2048)     W1 [IN][HID]
//   [2048, 2080)  b1 [HID]
//   [2080, 2592)  W2 [HID][CPAD]  (cols >= CLS stay zero)
//   [2592, 2608)  b2 [CPAD]
//   [2608]        loss (grads buffer only)
// ---------------------------------------------------------------------------

#define OFF_W1 0
#define OFF_B1 2048
#define OFF_W2 2080
#define OFF_B2 2592
#define OFF_LOSS 2608
#define NPARAM 2608

// ---------------------------------------------------------------------------
// shared device math for one 128-row chunk. All pointers are LDS arrays with
// the strides above. Each wave owns rows [wave*16, wave*16+16).
// Returns nothing; writes Hs/DLs/DHs, accumulates db/loss in LDS and the
// caller's dW accumulators in registers.
// ---------------------------------------------------------------------------

struct ChunkAcc {
  f32x4 dW1;        // this wave's dW1 tile (mt = wave>>1, nt = wave&1)
  f32x4 dW2;        // waves 0-1: dW2 tile (mt2 = wave)
};

__device__ __forceinline__ void chunk_fwd_bwd(
    const u16 (*Xs)[XS], u16 (*Hs)[HS], u16 (*DLs)[HS], u16 (*DHs)[HS],
    const u16 (*W1s)[HS], const u16 (*W2s)[W2S],
    const float* b1, const float* b2,
    const int* __restrict__ y, int row0, int B, float invBtot,
    float* db1_s, float* db2_s, float* loss_s, ChunkAcc& acc_io) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4, lr = l & 15;
  const int wrow = wave * 16;

  // fwd: H = relu(X @ W1 + b1)
  for (int nt = 0; nt < HID / 16; ++nt) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int ks = 0; ks < IN / 32; ++ks) {
      const bf16x8 a = *(const bf16x8*)&Xs[wrow + lr][ks * 32 + lg * 8];
      bf16x8 b;
      #pragma unroll
      for (int i = 0; i < 8; ++i) b[i] = (short)W1s[ks * 32 + lg * 8 + i][nt * 16 + lr];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      float h = acc[r] + b1[nt * 16 + lr];
      h = h > 0.f ? h : 0.f;
      Hs[wrow + lg * 4 + r][nt * 16 + lr] = f2bf(h);
    }
  }
  __syncthreads();   // Hs complete (bwd-W reads cross-wave rows)

  // logits + softmax + dlogits (intra-wave: own 16 rows)
  {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const bf16x8 a = *(const bf16x8*)&Hs[wrow + lr][lg * 8];
    bf16x8 b;
    #pragma unroll
    for (int i = 0; i < 8; ++i) b[i] = (short)W2s[lg * 8 + i][lr];
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);

    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wrow + lg * 4 + r;
      const bool valid_row = (row0 + row) < B;
      float logit = acc[r] + b2[lr];
      if (lr >= CLS) logit = -1e30f;
      float m = logit;
      #pragma unroll
      for (int d = 1; d < 16; d <<= 1) m = fmaxf(m, __shfl_xor(m, d, 64));
      const float e = (lr < CLS) ? __expf(logit - m) : 0.f;
      float s = e;
      #pragma unroll
      for (int d = 1; d < 16; d <<= 1) s += __shfl_xor(s, d, 64);
      const int label = valid_row ? y[row0 + row] : -1;
      const float dl = valid_row ? (e / s - (lr == label ? 1.f : 0.f)) * invBtot : 0.f;
      DLs[row][lr] = f2bf(dl);
      if (valid_row && lr == label) {
        atomicAdd(loss_s, -(logit - m - __logf(s)) * invBtot);
      }
      atomicAdd(&db2_s[lr], dl);
    }
  }

  // dH = dlogits @ W2^T (B-operand read transposed from W2s; k >= CPAD is 0)
  for (int nt = 0; nt < HID / 16; ++nt) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const bf16x8 a = *(const bf16x8*)&DLs[wrow + lr][lg * 8];
    bf16x8 b;
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int c = lg * 8 + i;
      b[i] = (c < CPAD) ? (short)W2s[nt * 16 + lr][c] : (short)0;
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wrow + lg * 4 + r;
      const float h = bf2f(Hs[row][nt * 16 + lr]);
      const float dh = h > 0.f ? acc[r] : 0.f;
      DHs[row][nt * 16 + lr] = f2bf(dh);
      atomicAdd(&db1_s[nt * 16 + lr], dh);
    }
  }
  __syncthreads();   // DHs/DLs complete for cross-wave bwd-W reads

  // dW1 = X^T @ dH: wave w owns tile (mt = w>>1, nt = w&1)
  {
    const int mt = wave >> 1, nt = wave & 1;
    f32x4 acc = acc_io.dW1;
    for (int ks = 0; ks < ROWS / 32; ++ks) {
      bf16x8 a, b;
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int k = ks * 32 + lg * 8 + i;        // batch row
        a[i] = (short)Xs[k][mt * 16 + lr];          // A[m][k] = X[k][m]
        b[i] = (short)DHs[k][nt * 16 + lr];
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    acc_io.dW1 = acc;
  }
  // dW2 = H^T @ dL: waves 0-1
  if (wave < 2) {
    f32x4 acc = acc_io.dW2;
    for (int ks = 0; ks < ROWS / 32; ++ks) {
      bf16x8 a, b;
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int k = ks * 32 + lg * 8 + i;
        a[i] = (short)Hs[k][wave * 16 + lr];
        b[i] = (short)DLs[k][lr];
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    acc_io.dW2 = acc;
  }
  __syncthreads();   // chunk arrays free for reuse
}

// cooperative X-chunk load (global bf16 -> LDS, zero-padded batch tail)
__device__ __forceinline__ void load_x_chunk(const u16* __restrict__ Xbf,
                                             u16 (*Xs)[XS], int row0, int B) {
  for (int i = threadIdx.x; i < ROWS * (IN / 8); i += BLOCK) {
    const int r = i / (IN / 8);
    const int c = (i % (IN / 8)) * 8;
    if (row0 + r < B) {
      *(bf16x8*)&Xs[r][c] = *(const bf16x8*)&Xbf[(long long)(row0 + r) * IN + c];
    } else {
      for (int k = 0; k < 8; ++k) Xs[r][c + k] = 0;
    }
  }
}

// zero DLs K-pad columns [CPAD, 32) once (rows never rewritten there)
__device__ __forceinline__ void zero_dl_pad(u16 (*DLs)[HS]) {
  for (int i = threadIdx.x; i < ROWS * CPAD; i += BLOCK) {
    const int r = i / CPAD, c = CPAD + (i % CPAD);
    DLs[r][c] = 0;
  }
}

// ---------------------------------------------------------------------------
// per-step kernel (multi-workgroup; DP path — grads via global atomics)
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(BLOCK)
mlp_step_kernel(const u16* __restrict__ Xbf, const int* __restrict__ y, int B,
                const u16* __restrict__ W1bf, const u16* __restrict__ W2bf,
                const float* __restrict__ master,
                float* __restrict__ grads, float invBtot) {
  __shared__ u16 Xs[ROWS][XS];
  __shared__ u16 Hs[ROWS][HS];
  __shared__ u16 DLs[ROWS][HS];
  __shared__ u16 DHs[ROWS][HS];
  __shared__ u16 W1s[IN][HS];
  __shared__ u16 W2s[HID][W2S];
  __shared__ float db1_s[HID], db2_s[CPAD], loss_s;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4, lr = l & 15;
  const int row0 = blockIdx.x * ROWS;

  if (tid < HID) db1_s[tid] = 0.f;
  if (tid < CPAD) db2_s[tid] = 0.f;
  if (tid == 0) loss_s = 0.f;
  zero_dl_pad(DLs);
  load_x_chunk(Xbf, Xs, row0, B);
  for (int i = tid; i < IN * (HID / 8); i += BLOCK) {
    const int r = i / (HID / 8), c = (i % (HID / 8)) * 8;
    *(bf16x8*)&W1s[r][c] = *(const bf16x8*)&W1bf[r * HID + c];
  }
  for (int i = tid; i < HID * (CPAD / 8); i += BLOCK) {
    const int r = i / (CPAD / 8), c = (i % (CPAD / 8)) * 8;
    *(bf16x8*)&W2s[r][c] = *(const bf16x8*)&W2bf[r * CPAD + c];
  }
  __syncthreads();

  ChunkAcc acc;
  acc.dW1 = (f32x4){0.f, 0.f, 0.f, 0.f};
  acc.dW2 = (f32x4){0.f, 0.f, 0.f, 0.f};
  chunk_fwd_bwd(Xs, Hs, DLs, DHs, W1s, W2s, master + OFF_B1, master + OFF_B2,
                y, row0, B, invBtot, db1_s, db2_s, &loss_s, acc);

  {
    const int mt = wave >> 1, nt = wave & 1;
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mrow = mt * 16 + lg * 4 + r;
      atomicAdd(&grads[OFF_W1 + mrow * HID + nt * 16 + lr], acc.dW1[r]);
    }
  }
  if (wave < 2) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int h = wave * 16 + lg * 4 + r;
      atomicAdd(&grads[OFF_W2 + h * CPAD + lr], acc.dW2[r]);
    }
  }
  if (tid < HID) atomicAdd(&grads[OFF_B1 + tid], db1_s[tid]);
  else if (tid < HID + CPAD) atomicAdd(&grads[OFF_B2 + tid - HID], db2_s[tid - HID]);
  else if (tid == HID + CPAD) atomicAdd(&grads[OFF_LOSS], loss_s);
}

// ---------------------------------------------------------------------------
// fused Adam (per-step DP path) — single block, device step counter
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
adam_step_kernel(float* __restrict__ master, u16* __restrict__ bfmirror,
                 const float* __restrict__ grads, float* __restrict__ m,
                 float* __restrict__ v, int* __restrict__ t_dev,
                 float lr, float beta1, float beta2, float eps) {
  __shared__ float corr1, corr2;
  if (threadIdx.x == 0) {
    const int t = ++(*t_dev);
    corr1 = 1.f / (1.f - __powf(beta1, (float)t));
    corr2 = 1.f / (1.f - __powf(beta2, (float)t));
  }
  __syncthreads();
  for (int i = threadIdx.x; i < NPARAM; i += 256) {
    const float g = grads[i];
    const float mi = beta1 * m[i] + (1.f - beta1) * g;
    const float vi = beta2 * v[i] + (1.f - beta2) * g * g;
    m[i] = mi;
    v[i] = vi;
    const float p = master[i] - lr * (mi * corr1) / (sqrtf(vi * corr2) + eps);
    master[i] = p;
    bfmirror[i] = f2bf(p);
  }
}

// ---------------------------------------------------------------------------
// THE single-GPU flagship: persistent multi-step kernel.
// ONE workgroup (8 waves). Dynamic LDS layout (16 B-aligned carves):
//   master_s [NPARAM] f32      | m_s [NPARAM] f32 | v_s [NPARAM] f32
//   W1s [IN][HS] bf16 | W2s [HID][W2S] bf16
//   Xs [ROWS][XS] | Hs/DLs/DHs [ROWS][HS] bf16
//   db1_s[HID] db2_s[CPAD] loss_s[4] f32
// Total ~87 KB — needs hipFuncAttributeMaxDynamicSharedMemorySize (set by
// the launcher); single workgroup, so 1-block-per-CU residency is free.
// ---------------------------------------------------------------------------

#define ALIGN16(x) (((x) + 15) & ~15)
#define LDS_MASTER 0
#define LDS_M      ALIGN16(LDS_MASTER + NPARAM * 4)
#define LDS_V      ALIGN16(LDS_M + NPARAM * 4)
#define LDS_W1S    ALIGN16(LDS_V + NPARAM * 4)
#define LDS_W2S    ALIGN16(LDS_W1S + IN * HS * 2)
#define LDS_XS     ALIGN16(LDS_W2S + HID * W2S * 2)
#define LDS_HS     ALIGN16(LDS_XS + ROWS * XS * 2)
#define LDS_DLS    ALIGN16(LDS_HS + ROWS * HS * 2)
#define LDS_DHS    ALIGN16(LDS_DLS + ROWS * HS * 2)
#define LDS_DB1    ALIGN16(LDS_DHS + ROWS * HS * 2)
#define LDS_DB2    ALIGN16(LDS_DB1 + HID * 4)
#define LDS_LOSS   ALIGN16(LDS_DB2 + CPAD * 4)
#define LDS_TOTAL  ALIGN16(LDS_LOSS + 16)

extern "C" __global__ void __launch_bounds__(BLOCK, 1)
mlp_train_steps_kernel(const u16* __restrict__ Xbf,  // [N][IN] staged bf16
                       const int* __restrict__ y,    // [N]
                       long long N, int B, int n_steps,
                       float* __restrict__ master,   // HBM fp32 in/out
                       u16* __restrict__ bfmirror,   // HBM bf16 out
                       float* __restrict__ m,        // HBM in/out
                       float* __restrict__ v,        // HBM in/out
                       int* __restrict__ t_dev,
                       float* __restrict__ loss_out, // last-step loss
                       float lr, float beta1, float beta2, float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* master_s = (float*)(smem + LDS_MASTER);
  float* m_s = (float*)(smem + LDS_M);
  float* v_s = (float*)(smem + LDS_V);
  u16 (*W1s)[HS] = (u16(*)[HS])(smem + LDS_W1S);
  u16 (*W2s)[W2S] = (u16(*)[W2S])(smem + LDS_W2S);
  u16 (*Xs)[XS] = (u16(*)[XS])(smem + LDS_XS);
  u16 (*Hs)[HS] = (u16(*)[HS])(smem + LDS_HS);
  u16 (*DLs)[HS] = (u16(*)[HS])(smem + LDS_DLS);
  u16 (*DHs)[HS] = (u16(*)[HS])(smem + LDS_DHS);
  float* db1_s = (float*)(smem + LDS_DB1);
  float* db2_s = (float*)(smem + LDS_DB2);
  float* loss_s = (float*)(smem + LDS_LOSS);

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4, ln = l & 15;   // ln: lane column (lr collides with the lr arg)

  // ---- load optimizer state + weights into LDS ------------------------------
  for (int i = tid; i < NPARAM; i += BLOCK) {
    const float p = master[i];
    master_s[i] = p;
    m_s[i] = m[i];
    v_s[i] = v[i];
    // fill bf16 weight images from the master (single source of truth)
    if (i < OFF_B1) W1s[i / HID][i % HID] = f2bf(p);
    else if (i >= OFF_W2 && i < OFF_B2) {
      const int j = i - OFF_W2;
      W2s[j / CPAD][j % CPAD] = f2bf(p);
    }
  }
  zero_dl_pad(DLs);
  const int t0 = *t_dev;
  // per-thread incremental bias-correction powers
  float b1t = __powf(beta1, (float)t0), b2t = __powf(beta2, (float)t0);
  __syncthreads();

  const int batches = (int)(N / B);
  const int chunks = B / ROWS;

  for (int s = 0; s < n_steps; ++s) {
    const long long base = (long long)(s % batches) * B;

    if (tid < HID) db1_s[tid] = 0.f;
    if (tid < CPAD) db2_s[tid] = 0.f;
    if (tid == 0) loss_s[0] = 0.f;

    ChunkAcc acc;
    acc.dW1 = (f32x4){0.f, 0.f, 0.f, 0.f};
    acc.dW2 = (f32x4){0.f, 0.f, 0.f, 0.f};
    const float invB = 1.f / (float)B;

    for (int ch = 0; ch < chunks; ++ch) {
      const int row0 = (int)base + ch * ROWS;
      load_x_chunk(Xbf + 0, Xs, row0, (int)N);   // rows always < N here
      __syncthreads();
      chunk_fwd_bwd(Xs, Hs, DLs, DHs, W1s, W2s, master_s + OFF_B1,
                    master_s + OFF_B2, y, row0, (int)N, invB,
                    db1_s, db2_s, loss_s, acc);
    }

    // ---- fused in-LDS Adam --------------------------------------------------
    b1t *= beta1;
    b2t *= beta2;
    const float corr1 = 1.f / (1.f - b1t);
    const float corr2 = 1.f / (1.f - b2t);

    // helper lambda-ish macro: update index i with grad g
    #define ADAM_UPD(i, g)                                                    \
      {                                                                       \
        const float mi = beta1 * m_s[i] + (1.f - beta1) * (g);                \
        const float vi = beta2 * v_s[i] + (1.f - beta2) * (g) * (g);          \
        m_s[i] = mi;                                                          \
        v_s[i] = vi;                                                          \
        master_s[i] = master_s[i] - lr * (mi * corr1) / (sqrtf(vi * corr2) + eps); \
      }

    {
      const int mt = wave >> 1, nt = wave & 1;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int mrow = mt * 16 + lg * 4 + r;
        const int idx = OFF_W1 + mrow * HID + nt * 16 + ln;
        ADAM_UPD(idx, acc.dW1[r]);
        W1s[mrow][nt * 16 + ln] = f2bf(master_s[idx]);
      }
    }
    if (wave < 2) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int h = wave * 16 + lg * 4 + r;
        const int idx = OFF_W2 + h * CPAD + ln;
        ADAM_UPD(idx, acc.dW2[r]);
        W2s[h][ln] = f2bf(master_s[idx]);
      }
    }
    if (wave == 2) {
      // biases: lanes 0..31 -> b1, lanes 32..47 -> b2
      if (l < HID) {
        ADAM_UPD(OFF_B1 + l, db1_s[l]);
      } else if (l < HID + CPAD) {
        ADAM_UPD(OFF_B2 + (l - HID), db2_s[l - HID]);
      }
    }
    #undef ADAM_UPD
    __syncthreads();   // weights updated before next step's fwd
  }

  // ---- write state back to HBM ----------------------------------------------
  if (tid == 0) {
    *t_dev = t0 + n_steps;
    *loss_out = loss_s[0];
  }
  for (int i = tid; i < NPARAM; i += BLOCK) {
    master[i] = master_s[i];
    m[i] = m_s[i];
    v[i] = v_s[i];
    bfmirror[i] = f2bf(master_s[i]);
  }
}

// ---------------------------------------------------------------------------
// fused predict: standardize + fwd + argmax (serving hot path)
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(BLOCK)
mlp_predict_kernel(const float* __restrict__ X, int B,
                   const float* __restrict__ mean,
                   const float* __restrict__ invstd,
                   const u16* __restrict__ W1bf, const u16* __restrict__ W2bf,
                   const float* __restrict__ master,
                   int* __restrict__ preds,
                   float* __restrict__ probs /* optional [B][CLS] */) {
  __shared__ u16 Xs[ROWS][XS];
  __shared__ u16 Hs[ROWS][HS];
  __shared__ u16 W1s[IN][HS];
  __shared__ u16 W2s[HID][W2S];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4, lr = l & 15;
  const int row0 = blockIdx.x * ROWS;
  const int wrow = wave * 16;

  for (int i = tid; i < ROWS * IN; i += BLOCK) {
    const int r = i / IN, c = i % IN;
    const float v =
        (row0 + r < B) ? (X[(long long)(row0 + r) * IN + c] - mean[c]) * invstd[c] : 0.f;
    Xs[r][c] = f2bf(v);
  }
  for (int i = tid; i < IN * (HID / 8); i += BLOCK) {
    const int r = i / (HID / 8), c = (i % (HID / 8)) * 8;
    *(bf16x8*)&W1s[r][c] = *(const bf16x8*)&W1bf[r * HID + c];
  }
  for (int i = tid; i < HID * (CPAD / 8); i += BLOCK) {
    const int r = i / (CPAD / 8), c = (i % (CPAD / 8)) * 8;
    *(bf16x8*)&W2s[r][c] = *(const bf16x8*)&W2bf[r * CPAD + c];
  }
  __syncthreads();

  const float* b1 = master + OFF_B1;
  const float* b2 = master + OFF_B2;

  for (int nt = 0; nt < HID / 16; ++nt) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int ks = 0; ks < IN / 32; ++ks) {
      const bf16x8 a = *(const bf16x8*)&Xs[wrow + lr][ks * 32 + lg * 8];
      bf16x8 b;
      #pragma unroll
      for (int i = 0; i < 8; ++i) b[i] = (short)W1s[ks * 32 + lg * 8 + i][nt * 16 + lr];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      float h = acc[r] + b1[nt * 16 + lr];
      Hs[wrow + lg * 4 + r][nt * 16 + lr] = f2bf(h > 0.f ? h : 0.f);
    }
  }
  __syncthreads();

  {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const bf16x8 a = *(const bf16x8*)&Hs[wrow + lr][lg * 8];
    bf16x8 b;
    #pragma unroll
    for (int i = 0; i < 8; ++i) b[i] = (short)W2s[lg * 8 + i][lr];
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);

    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wrow + lg * 4 + r;
      float logit = acc[r] + b2[lr];
      if (lr >= CLS) logit = -1e30f;
      // argmax across the row's 16 lanes; ties pick the lowest column
      float best = logit;
      int bcol = lr;
      #pragma unroll
      for (int d = 1; d < 16; d <<= 1) {
        const float ov = __shfl_xor(best, d, 64);
        const int oc = __shfl_xor(bcol, d, 64);
        if (ov > best || (ov == best && oc < bcol)) { best = ov; bcol = oc; }
      }
      if (lr == 0 && row0 + row < B) preds[row0 + row] = bcol;
      if (probs != nullptr) {
        // shuffle reductions run with ALL lanes active; only the write is
        // guarded (an inactive lane's shfl result is undefined)
        const float e = __expf(logit - best);   // best == row max
        float s = e;
        #pragma unroll
        for (int d = 1; d < 16; d <<= 1) s += __shfl_xor(s, d, 64);
        if (lr < CLS && row0 + row < B) {
          probs[(long long)(row0 + row) * CLS + lr] = e / s;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host launchers (extern "C"; stream-ordered, capture-safe)
// ---------------------------------------------------------------------------

extern "C" {

void launch_standardize_fit(const float* X, long long N, int D, float* mean,
                            float* invstd, float eps, hipStream_t stream) {
  hipLaunchKernelGGL(standardize_fit_kernel, dim3(D), dim3(256), 0, stream,
                     X, N, D, mean, invstd, eps);
}

void launch_standardize_apply(const float* X, long long N, int D,
                              const float* mean, const float* invstd,
                              unsigned short* out, hipStream_t stream) {
  const long long n = N * D;
  int blocks = (int)((n + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(standardize_apply_kernel, dim3(blocks), dim3(256), 0, stream,
                     X, n, D, mean, invstd, out);
}

void launch_mlp_step(const unsigned short* Xbf, const int* y, int B,
                     const unsigned short* W1bf, const unsigned short* W2bf,
                     const float* master, float* grads, float invBtot,
                     hipStream_t stream) {
  const int blocks = (B + ROWS - 1) / ROWS;
  hipLaunchKernelGGL(mlp_step_kernel, dim3(blocks), dim3(BLOCK), 0, stream,
                     Xbf, y, B, W1bf, W2bf, master, grads, invBtot);
}

int launch_mlp_train_steps(const unsigned short* Xbf, const int* y, long long N,
                           int B, int n_steps, float* master,
                           unsigned short* bfmirror, float* m, float* v,
                           int* t_dev, float* loss_out, float lr, float beta1,
                           float beta2, float eps, hipStream_t stream) {
  if (B % ROWS != 0 || N % B != 0) return -1;   // caller falls back
  static int lds_ok = 0;
  if (!lds_ok) {
    hipError_t err = hipFuncSetAttribute(
        (const void*)mlp_train_steps_kernel,
        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_TOTAL);
    if (err != hipSuccess) return -2;
    lds_ok = 1;
  }
  hipLaunchKernelGGL(mlp_train_steps_kernel, dim3(1), dim3(BLOCK), LDS_TOTAL,
                     stream, Xbf, y, N, B, n_steps, master, bfmirror, m, v,
                     t_dev, loss_out, lr, beta1, beta2, eps);
  return 0;
}

void launch_mlp_predict(const float* X, int B, const float* mean,
                        const float* invstd, const unsigned short* W1bf,
                        const unsigned short* W2bf, const float* master,
                        int* preds, float* probs, hipStream_t stream) {
  const int blocks = (B + ROWS - 1) / ROWS;
  hipLaunchKernelGGL(mlp_predict_kernel, dim3(blocks), dim3(BLOCK), 0, stream,
                     X, B, mean, invstd, W1bf, W2bf, master, preds, probs);
}

void launch_adam_step(float* master, unsigned short* bfmirror, const float* grads,
                      float* m, float* v, int* t_dev, float lr, float beta1,
                      float beta2, float eps, hipStream_t stream) {
  hipLaunchKernelGGL(adam_step_kernel, dim3(1), dim3(256), 0, stream,
                     master, bfmirror, grads, m, v, t_dev, lr, beta1, beta2, eps);
}

}  // extern "C"
