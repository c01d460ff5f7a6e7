// CDNA4 (gfx950 / MI355X) kernels for the default tabular hot path.
//
// Implements the kernel set of SURVEY.md §2c (net-new designs; the
// reference unionai-oss/unionml is pure Python and has no kernels):
//   - standardize_fit / standardize_apply   : per-column (x-mean)*invstd, fp32 -> bf16
//   - mlp_step                              : fused fwd+bwd of the digits MLP
//                                             (IN=64 -> HID=32 relu -> CLS=10 softmax/xent)
//                                             producing fp32 grads + loss in ONE launch
//   - mlp_predict                           : fused standardize + fwd + argmax
//   - adam_step                             : single-block fused Adam on the flat
//                                             fp32 master params, emitting the bf16
//                                             compute mirror
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//   * wave64; block = 8 waves (512 threads); each wave owns 16 rows of the batch.
//   * All GEMM-shaped work on MFMA: __builtin_amdgcn_mfma_f32_16x16x32_bf16
//     (gfx950 2xK form), fp32 accumulate. Fragment mapping:
//       A[16x32]:  lane l holds A[l&15][(l>>4)*8 + i], i = 0..7
//       B[32x16]:  lane l holds B[(l>>4)*8 + i][l&15]
//       C/D[16x16]: lane l, reg r holds D[(l>>4)*4 + r][l&15]
//     (verified on hardware by tests/test_gpu_kernels.py with asymmetric operands).
//   * Whole batch tile staged in LDS; rows padded +8 bf16 (16 B) so the
//     ds_read_b128 A-fragment reads hit 16 distinct bank slots (guide §6 G4).
//   * Weight-gradient tiles (X^T dH, H^T dL) accumulate in AGPRs over the
//     row tile, then one fp32 atomicAdd per element merges across workgroups.
//   * Every launch is stream-ordered and hipGraph-capturable (no mallocs,
//     no syncs — guide Guideline 9). Adam reads its step counter from a
//     device int so bias correction stays correct under graph replay.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define IN 64
#define HID 32
#define CLS 10
#define CPAD 16          // CLS padded to one MFMA tile
#define ROWS 128         // batch rows per workgroup
#define WAVES 8
#define BLOCK (WAVES * 64)

// LDS row strides, padded so every ds_read_b128 A-fragment address stays
// 16 B aligned (row_stride_bytes % 16 == 0 — guide §6 G17) while breaking
// the power-of-2 bank pattern (guide §6 G4).
#define XS 72            // IN + 8   (144 B rows)
#define HS 48            // HID + 16 (96 B rows)
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
// standardize: column mean / inverse std (Welford-free two-sum, fp64 accum)
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
  // grid-stride over elements; D divides 64 so col = idx % D stays cheap
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n_elems;
       i += (long long)gridDim.x * blockDim.x) {
    const int col = (int)(i % D);
    out[i] = f2bf((X[i] - mean[col]) * invstd[col]);
  }
}

// ---------------------------------------------------------------------------
// fused MLP train step: fwd + xent bwd + weight grads, one launch
//
// grads/master flat layout (floats):
//   [0, 2048)        W1   [IN][HID]
//   [2048, 2080)     b1   [HID]
//   [2080, 2592)     W2   [HID][CPAD]   (cols >= CLS stay zero)
//   [2592, 2608)     b2   [CPAD]
//   [2608]           loss (grads buffer only)
// ---------------------------------------------------------------------------

#define OFF_W1 0
#define OFF_B1 2048
#define OFF_W2 2080
#define OFF_B2 2592
#define OFF_LOSS 2608
#define NPARAM 2608

__device__ __forceinline__ bf16x8 lds_frag_a(const u16* base) {
  // 8 contiguous bf16 at a 16B-aligned LDS address -> ds_read_b128
  return *(const bf16x8*)base;
}

extern "C" __global__ void __launch_bounds__(BLOCK)
mlp_step_kernel(const u16* __restrict__ Xbf,     // [B][IN] standardized bf16
                const int* __restrict__ y,       // [B] labels
                int B,
                const u16* __restrict__ W1bf,    // [IN][HID]
                const u16* __restrict__ W2bf,    // [HID][CPAD]
                const float* __restrict__ master,// biases read at OFF_B1/OFF_B2
                float* __restrict__ grads,       // flat, pre-zeroed, +loss
                float invBtot) {
  __shared__ u16 Xs[ROWS][XS];
  __shared__ u16 Hs[ROWS][HS];
  __shared__ u16 DLs[ROWS][HS];   // dlogits padded to K=32 (cols CPAD..31 zero)
  __shared__ u16 DHs[ROWS][HS];
  __shared__ u16 W1s[IN][HS];
  __shared__ u16 W2s[HID][W2S];
  __shared__ u16 W2Ts[32][HS];    // W2^T, k padded to 32
  __shared__ float db1_s[HID], db2_s[CPAD], loss_s;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4;           // 16-lane group 0..3
  const int lr = l & 15;
  const int row0 = blockIdx.x * ROWS;   // this WG's batch offset
  const int wrow = wave * 16;           // this wave's row block inside the tile

  // ---- zero LDS accumulators & padded regions -------------------------------
  if (tid < HID) db1_s[tid] = 0.f;
  if (tid < CPAD) db2_s[tid] = 0.f;
  if (tid == 0) loss_s = 0.f;
  // DLs upper K half must be zero for the dH MFMA
  for (int i = tid; i < ROWS * (HS - CPAD) / 8; i += BLOCK) {
    // zero cols [CPAD, HS) row by row (8 cols per thread-slot)
    const int r = i / ((HS - CPAD) / 8);
    const int c = CPAD + (i % ((HS - CPAD) / 8)) * 8;
    for (int k = 0; k < 8; ++k) DLs[r][c + k] = 0;
  }
  // W2Ts rows CLS..31 zero
  for (int i = tid; i < 32 * HID; i += BLOCK) {
    const int k = i / HID, n = i % HID;
    W2Ts[k][n] = 0;
  }

  // ---- cooperative loads ----------------------------------------------------
  // X rows (guard the batch tail with zero rows)
  for (int i = tid; i < ROWS * (IN / 8); i += BLOCK) {
    const int r = i / (IN / 8);
    const int c = (i % (IN / 8)) * 8;
    if (row0 + r < B) {
      *(bf16x8*)&Xs[r][c] = *(const bf16x8*)&Xbf[(long long)(row0 + r) * IN + c];
    } else {
      for (int k = 0; k < 8; ++k) Xs[r][c + k] = 0;
    }
  }
  for (int i = tid; i < IN * (HID / 8); i += BLOCK) {
    const int r = i / (HID / 8);
    const int c = (i % (HID / 8)) * 8;
    *(bf16x8*)&W1s[r][c] = *(const bf16x8*)&W1bf[r * HID + c];
  }
  for (int i = tid; i < HID * (CPAD / 8); i += BLOCK) {
    const int r = i / (CPAD / 8);
    const int c = (i % (CPAD / 8)) * 8;
    *(bf16x8*)&W2s[r][c] = *(const bf16x8*)&W2bf[r * CPAD + c];
  }
  __syncthreads();
  // W2^T fill (after W2s is resident)
  for (int i = tid; i < CLS * HID; i += BLOCK) {
    const int c = i / HID, h = i % HID;
    W2Ts[c][h] = W2s[h][c];
  }
  __syncthreads();

  // ---- forward: H = relu(X @ W1 + b1) --------------------------------------
  const float* b1 = master + OFF_B1;
  const float* b2 = master + OFF_B2;

  for (int nt = 0; nt < HID / 16; ++nt) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int ks = 0; ks < IN / 32; ++ks) {
      const bf16x8 a = lds_frag_a(&Xs[wrow + lr][ks * 32 + lg * 8]);
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
  __syncthreads();

  // ---- logits + softmax + dlogits -------------------------------------------
  {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const bf16x8 a = lds_frag_a(&Hs[wrow + lr][lg * 8]);
    bf16x8 b;
    #pragma unroll
    for (int i = 0; i < 8; ++i) b[i] = (short)W2s[lg * 8 + i][lr];
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);

    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wrow + lg * 4 + r;              // row within tile
      const bool valid_row = (row0 + row) < B;
      float logit = acc[r] + b2[lr];
      if (lr >= CLS) logit = -1e30f;
      // row max across the 16 lanes holding this row's columns
      float m = logit;
      #pragma unroll
      for (int d = 1; d < 16; d <<= 1) m = fmaxf(m, __shfl_xor(m, d, 64));
      const float e = (lr < CLS) ? __expf(logit - m) : 0.f;
      float s = e;
      #pragma unroll
      for (int d = 1; d < 16; d <<= 1) s += __shfl_xor(s, d, 64);
      const int label = valid_row ? y[row0 + row] : -1;
      const float p = e / s;
      float dl = valid_row ? (p - (lr == label ? 1.f : 0.f)) * invBtot : 0.f;
      DLs[row][lr] = f2bf(dl);
      if (valid_row && lr == label) {
        atomicAdd(&loss_s, -(logit - m - __logf(s)) * invBtot);
      }
      atomicAdd(&db2_s[lr], dl);
    }
  }
  __syncthreads();

  // ---- dH = dlogits @ W2^T, relu mask ---------------------------------------
  for (int nt = 0; nt < HID / 16; ++nt) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const bf16x8 a = lds_frag_a(&DLs[wrow + lr][lg * 8]);
    bf16x8 b;
    #pragma unroll
    for (int i = 0; i < 8; ++i) b[i] = (short)W2Ts[lg * 8 + i][nt * 16 + lr];
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wrow + lg * 4 + r;
      const float h = bf2f(Hs[row][nt * 16 + lr]);
      float dh = h > 0.f ? acc[r] : 0.f;
      DHs[row][nt * 16 + lr] = f2bf(dh);
      atomicAdd(&db1_s[nt * 16 + lr], dh);
    }
  }
  __syncthreads();

  // ---- weight grads: dW1 = X^T @ dH (8 tiles), dW2 = H^T @ dL (2 tiles) -----
  // wave w owns dW1 tile w (mt = w>>1, nt = w&1); waves 0-1 also own dW2 tiles.
  {
    const int mt = wave >> 1, nt = wave & 1;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int ks = 0; ks < ROWS / 32; ++ks) {
      bf16x8 a, b;
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int k = ks * 32 + lg * 8 + i;        // batch row
        a[i] = (short)Xs[k][mt * 16 + lr];          // A[m][k] = X[k][m]
        b[i] = (short)DHs[k][nt * 16 + lr];         // B[k][n] = dH[k][n]
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = mt * 16 + lg * 4 + r;           // input-feature index
      atomicAdd(&grads[OFF_W1 + m * HID + nt * 16 + lr], acc[r]);
    }
  }
  if (wave < 2) {
    const int mt = wave;                            // hidden-unit tile
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int ks = 0; ks < ROWS / 32; ++ks) {
      bf16x8 a, b;
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int k = ks * 32 + lg * 8 + i;
        a[i] = (short)Hs[k][mt * 16 + lr];          // A[m][k] = H[k][m]
        b[i] = (short)DLs[k][lr];                   // B[k][n] = dL[k][n]
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int h = mt * 16 + lg * 4 + r;
      atomicAdd(&grads[OFF_W2 + h * CPAD + lr], acc[r]);
    }
  }

  // ---- bias grads + loss to global ------------------------------------------
  __syncthreads();
  if (tid < HID) atomicAdd(&grads[OFF_B1 + tid], db1_s[tid]);
  else if (tid < HID + CPAD) atomicAdd(&grads[OFF_B2 + tid - HID], db2_s[tid - HID]);
  else if (tid == HID + CPAD) atomicAdd(&grads[OFF_LOSS], loss_s);
}

// ---------------------------------------------------------------------------
// fused predict: standardize + fwd + argmax (serving hot path)
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(BLOCK)
mlp_predict_kernel(const float* __restrict__ X,   // [B][IN] raw fp32
                   int B,
                   const float* __restrict__ mean,
                   const float* __restrict__ invstd,
                   const u16* __restrict__ W1bf,
                   const u16* __restrict__ W2bf,
                   const float* __restrict__ master,
                   int* __restrict__ preds,
                   float* __restrict__ probs /* optional [B][CLS], may be null */) {
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

  // standardize on load (fp32 -> bf16)
  for (int i = tid; i < ROWS * IN; i += BLOCK) {
    const int r = i / IN, c = i % IN;
    float v = (row0 + r < B) ? (X[(long long)(row0 + r) * IN + c] - mean[c]) * invstd[c] : 0.f;
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
      const bf16x8 a = lds_frag_a(&Xs[wrow + lr][ks * 32 + lg * 8]);
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
    const bf16x8 a = lds_frag_a(&Hs[wrow + lr][lg * 8]);
    bf16x8 b;
    #pragma unroll
    for (int i = 0; i < 8; ++i) b[i] = (short)W2s[lg * 8 + i][lr];
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);

    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wrow + lg * 4 + r;
      float logit = acc[r] + b2[lr];
      if (lr >= CLS) logit = -1e30f;
      // argmax across the row's 16 lanes: pack (logit, 15-col) so ties pick
      // the LOWEST column, matching torch.argmax
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
        // shuffle reductions must run with ALL lanes active (an inactive
        // lane's shfl result is undefined) — only the write is guarded
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
// fused Adam — single block (NPARAM = 2608 floats), graph-replay safe:
// the step counter lives in device memory and the kernel ticks it itself.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
adam_step_kernel(float* __restrict__ master,
                 u16* __restrict__ bfmirror,
                 const float* __restrict__ grads,
                 float* __restrict__ m,
                 float* __restrict__ v,
                 int* __restrict__ t_dev,
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
  if (blocks > 2048) blocks = 2048;   // grid-stride beyond (guide G11)
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
