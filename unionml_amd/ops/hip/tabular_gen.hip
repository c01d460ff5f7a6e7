// CDNA4 (gfx950) GENERALIZED tabular MLP kernels: any (in_features,
// hidden, classes) geometry, not just the tuned 64x32x10 flagship of
// tabular_kernels.hip.
//
// Design (MI355X-first, not a port — the reference unionai-oss/unionml
// has no kernels; behavior mirrors its pytorch quickstart trainer,
// tests/integration/pytorch_app/quickstart.py:14-70):
//
//   * Template parameters are ONLY <HID, RT> (hidden width, rows per
//     workgroup). The input width is RUNTIME: it appears solely in
//     k-tile counts and global strides, so four instantiations
//     (HID 32/64/128/256) cover every geometry. Odd widths are
//     zero-padded by the Python layer (exact math: zero-init + zero
//     gradient keeps Adam at exactly zero for pad params).
//   * The input dimension streams through LDS in TK=64 k-tiles (a
//     784-feature MNIST row set no longer has to fit in LDS whole);
//     weights live in packed global "images" in kernel-friendly
//     layouts (W1T row-major over K, W2s K-padded, W2T) kept in sync
//     by the in-kernel Adam phase — the same packed-image scheme the
//     specialized kernel's wimg uses.
//   * The training step is TWO kernels: the slab-producing MFMA
//     fwd/bwd (grid = batch rows / rows-per-WG, plain per-WG stores,
//     no atomics) and a wide-grid reduce+Adam kernel whose launch
//     boundary doubles as the inter-workgroup barrier — at generalized
//     nparam sizes an in-grid reduce is bandwidth-starved (memory
//     parallelism scales with resident CUs), so the specialized
//     kernel's G16 epoch-tag handshake is deliberately NOT used here
//     (A/B trail: profiles/r02_gen_kernel_stats.md). MFMA
//     __builtin_amdgcn_mfma_f32_16x16x32_bf16 with the
//     hardware-verified fragment mapping documented in
//     tabular_kernels.hip:34-46. Classes are runtime (<= 32, one or
//     two MFMA tiles): the wave-shuffle softmax masks c >= cls and
//     folds across the register-resident class-tile axis.
//
// All launches are stream-ordered and hipGraph-capturable; the Adam
// step counter lives in device memory so bias correction is exact
// under graph replay.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef unsigned short u16;

namespace gen {

__device__ __forceinline__ float bf2f(u16 v) {
  union { float f; unsigned u; } c;
  c.u = ((unsigned)v) << 16;
  return c.f;
}

__device__ __forceinline__ u16 f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;  // round-to-nearest-even
  return (u16)(c.u >> 16);
}

__device__ __forceinline__ float fast_rcp(float x) {
  return __builtin_amdgcn_rcpf(x);
}

constexpr int A16(int x) { return (x + 15) & ~15; }
constexpr int IMAX(int a, int b) { return a > b ? a : b; }

// Geometry: LDS layout + tile bookkeeping, all from <HID, RT, CP>.
template <int HID_, int RT_, int CP_ = 16>
struct GG {
  static constexpr int HID = HID_;
  static constexpr int RT = RT_;        // batch rows per workgroup
  // k-tile over the input width; the widest hidden's W1T tile would
  // not fit double-buffered at TK=64, so it streams narrower tiles
  // (measured -14% at the hid=256 instantiation)
  static constexpr int TK = (HID_ >= 256) ? 32 : 64;
  static constexpr int CPAD = CP_;      // classes padded to MFMA tiles (16/32)
  static constexpr int NCT = CPAD / 16; // class tiles in the fwd2 head
  static_assert(CPAD == 16 || CPAD == 32, "classifier head: 16 or 32 classes");
  static constexpr int WAVES = 8;
  static constexpr int BLOCK = WAVES * 64;

  // LDS row strides (u16 elems), +8 keeps b128 alignment w/ distinct
  // bank slots per row (same scheme as tabular_kernels.hip:60-63)
  static constexpr int XSS = TK + 8;    // Xs tile   [RT][XSS]
  static constexpr int XTS = RT + 8;    // XT tile   [TK][XTS]   (shares xbuf)
  static constexpr int W1S = TK + 8;    // W1T tile  [HID][W1S]
  static constexpr int HSS = HID + 8;   // Hs        [RT][HSS]
  static constexpr int HTS = RT + 8;    // HT        [HID][HTS]
  static constexpr int DLS = 40;        // DLs       [RT][40]   (K-pad to 32)
  static constexpr int DTS = RT + 8;    // DLT       [CPAD][DTS]
  static constexpr int DHS = RT + 8;    // DHT       [HID][DHS]
  static constexpr int W2S = 40;        // W2s       [HID][40]  (K-pad to 32)
  static constexpr int W2T = HID + 8;   // W2T       [CPAD][W2T]

  // streamed-tile buffer sizes (u16 elems, ONE buffer each)
  static constexpr int XBUF1 = IMAX(RT * XSS, TK * XTS);
  static constexpr int W1B1 = HID * W1S;

  static constexpr int layout_total(int nxb) {
    int o = A16(nxb * XBUF1 * 2);
    o = A16(o + nxb * W1B1 * 2);
    o = A16(o + RT * HSS * 2);
    o = A16(o + HID * HTS * 2);
    o = A16(o + RT * DLS * 2);
    o = A16(o + CPAD * DTS * 2);
    o = A16(o + HID * DHS * 2);
    o = A16(o + HID * W2S * 2);
    o = A16(o + CPAD * W2T * 2);
    o = A16(o + HID * 4);
    o = A16(o + CPAD * 4);
    o = A16(o + 16);
    o = A16(o + HID * 4);
    return A16(o + CPAD * 4);
  }

  // double-buffer the k-streamed tiles (Xs/W1T in fwd1, XT in dW1)
  // whenever the carve still fits in the CU's 160 KB: the next tile's
  // cooperative load then overlaps the current tile's MFMAs, hiding
  // the L2/HBM round trip that otherwise serializes each k step
  static constexpr bool DB = layout_total(2) <= 160 * 1024;
  static constexpr int NXB = DB ? 2 : 1;

  static constexpr int O_XB = 0;
  static constexpr int O_W1 = A16(O_XB + NXB * XBUF1 * 2);
  static constexpr int O_HS = A16(O_W1 + NXB * W1B1 * 2);
  static constexpr int O_HT = A16(O_HS + RT * HSS * 2);
  static constexpr int O_DL = A16(O_HT + HID * HTS * 2);
  static constexpr int O_DT = A16(O_DL + RT * DLS * 2);
  static constexpr int O_DH = A16(O_DT + CPAD * DTS * 2);
  static constexpr int O_2S = A16(O_DH + HID * DHS * 2);
  static constexpr int O_2T = A16(O_2S + HID * W2S * 2);
  static constexpr int O_B1G = A16(O_2T + CPAD * W2T * 2);   // db1 [HID] f32
  static constexpr int O_B2G = A16(O_B1G + HID * 4);         // db2 [CPAD] f32
  static constexpr int O_LS = A16(O_B2G + CPAD * 4);         // loss slots (4 f32)
  static constexpr int O_B1 = A16(O_LS + 16);                // b1 stage [HID]
  static constexpr int O_B2 = A16(O_B1 + HID * 4);           // b2 stage [CPAD]
  static constexpr int TOTAL = A16(O_B2 + CPAD * 4);
  static_assert(TOTAL == layout_total(NXB), "layout helper out of sync");

  static constexpr int NT1 = (HID / 16) * (RT / 16);  // fwd1 / dH D tiles
  static constexpr int SLOTS = (NT1 + WAVES - 1) / WAVES;

  static_assert(HID % 16 == 0, "HID must be a multiple of 16");
  static_assert(RT % 32 == 0, "RT must be a multiple of 32");
  static_assert(TOTAL <= 160 * 1024, "LDS carve exceeds 160 KB");

  char* smem_;
  u16 (*Hs)[HSS];
  u16 (*HT)[HTS];
  u16 (*DLs)[DLS];
  u16 (*DLT)[DTS];
  u16 (*DHT)[DHS];
  u16 (*W2s)[W2S];
  u16 (*W2Tt)[W2T];
  float* db1;
  float* db2;
  float* loss;
  float* b1s;
  float* b2s;

  __device__ __forceinline__ u16 (*Xs(int b))[XSS] {
    return (u16(*)[XSS])(smem_ + O_XB + b * XBUF1 * 2);
  }
  __device__ __forceinline__ u16 (*XT(int b))[XTS] {
    return (u16(*)[XTS])(smem_ + O_XB + b * XBUF1 * 2);
  }
  __device__ __forceinline__ u16 (*W1T(int b))[W1S] {
    return (u16(*)[W1S])(smem_ + O_W1 + b * W1B1 * 2);
  }

  __device__ __forceinline__ void carve(char* smem) {
    smem_ = smem;
    Hs = (u16(*)[HSS])(smem + O_HS);
    HT = (u16(*)[HTS])(smem + O_HT);
    DLs = (u16(*)[DLS])(smem + O_DL);
    DLT = (u16(*)[DTS])(smem + O_DT);
    DHT = (u16(*)[DHS])(smem + O_DH);
    W2s = (u16(*)[W2S])(smem + O_2S);
    W2Tt = (u16(*)[W2T])(smem + O_2T);
    db1 = (float*)(smem + O_B1G);
    db2 = (float*)(smem + O_B2G);
    loss = (float*)(smem + O_LS);
    b1s = (float*)(smem + O_B1);
    b2s = (float*)(smem + O_B2);
  }
};

// packed global weight-image offsets (bf16 elems): W1Tg [HID][inp] then
// W2sg [HID][32] (cols >= 16 zero) then W2Tg [16][HID]
__device__ __forceinline__ int wimg_w2s_off(int hid, int inp) { return hid * inp; }
__device__ __forceinline__ int wimg_w2t_off(int hid, int inp) {
  return hid * inp + hid * 32;
}

// Adam-phase image maintenance: param i -> packed-image stores
template <int HID, int CP>
__device__ __forceinline__ void wimg_write_gen(u16* __restrict__ wimg, int inp,
                                               int off_b1, int off_w2, int off_b2,
                                               int i, u16 wb) {
  if (i < off_b1) {                       // W1: i = in*HID + h
    const int in = i / HID;
    const int h = i % HID;
    wimg[h * inp + in] = wb;
  } else if (i >= off_w2 && i < off_b2) { // W2: j = h*CP + c
    const int j = i - off_w2;
    const int h = j / CP;
    const int c = j % CP;
    wimg[wimg_w2s_off(HID, inp) + h * 32 + c] = wb;
    wimg[wimg_w2t_off(HID, inp) + c * HID + h] = wb;
  }
}

// ---------------------------------------------------------------------------
// cooperative LDS loaders (all BLOCK threads)
// ---------------------------------------------------------------------------

// Xs tile: rows of this WG's chunk, k-columns [k0, k0+kv), zero tail
template <typename G>
__device__ __forceinline__ void load_xs_tile(G& L, int buf,
                                             const u16* __restrict__ Xbf,
                                             int inp, long long row0,
                                             long long Nvalid, int k0, int kv) {
  u16 (*Xs)[G::XSS] = L.Xs(buf);
  for (int i = threadIdx.x; i < G::RT * (G::TK / 8); i += G::BLOCK) {
    const int r = i / (G::TK / 8);
    const int jg = (i % (G::TK / 8)) * 8;
    bf16x8 v = (bf16x8){0, 0, 0, 0, 0, 0, 0, 0};
    if (row0 + r < Nvalid && jg < kv) {
      v = *(const bf16x8*)&Xbf[(row0 + r) * inp + k0 + jg];
    }
    *(bf16x8*)&Xs[r][jg] = v;
  }
}

// XT tile: transposed image of the same chunk/k-tile (dW1's B operand)
template <typename G>
__device__ __forceinline__ void load_xt_tile(G& L, int buf,
                                             const u16* __restrict__ Xbf,
                                             int inp, long long row0,
                                             long long Nvalid, int k0, int kv) {
  u16 (*XT)[G::XTS] = L.XT(buf);
  for (int i = threadIdx.x; i < G::RT * (G::TK / 8); i += G::BLOCK) {
    const int r = i / (G::TK / 8);
    const int jg = (i % (G::TK / 8)) * 8;
    bf16x8 v = (bf16x8){0, 0, 0, 0, 0, 0, 0, 0};
    if (row0 + r < Nvalid && jg < kv) {
      v = *(const bf16x8*)&Xbf[(row0 + r) * inp + k0 + jg];
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) XT[jg + j][r] = (u16)v[j];
  }
}

// W1T k-tile from the packed global image (row-major [HID][inp])
template <typename G>
__device__ __forceinline__ void load_w1t_tile(G& L, int buf,
                                              const u16* __restrict__ wimg,
                                              int inp, int k0, int kv) {
  u16 (*W1T)[G::W1S] = L.W1T(buf);
  for (int i = threadIdx.x; i < G::HID * (G::TK / 8); i += G::BLOCK) {
    const int h = i / (G::TK / 8);
    const int jg = (i % (G::TK / 8)) * 8;
    bf16x8 v = (bf16x8){0, 0, 0, 0, 0, 0, 0, 0};
    if (jg < kv) v = *(const bf16x8*)&wimg[h * inp + k0 + jg];
    *(bf16x8*)&W1T[h][jg] = v;
  }
}

// whole W2 images (small): W2s [HID][40] from [HID][32]; W2T [16][HID+8]
template <typename G>
__device__ __forceinline__ void load_w2_images(const G& L, const u16* __restrict__ wimg,
                                               int inp) {
  const u16* w2s = wimg + wimg_w2s_off(G::HID, inp);
  const u16* w2t = wimg + wimg_w2t_off(G::HID, inp);
  for (int i = threadIdx.x; i < G::HID * 4; i += G::BLOCK) {
    const int h = i / 4, jg = (i % 4) * 8;
    *(bf16x8*)&L.W2s[h][jg] = *(const bf16x8*)&w2s[h * 32 + jg];
  }
  for (int i = threadIdx.x; i < G::CPAD * (G::HID / 8); i += G::BLOCK) {
    const int c = i / (G::HID / 8), jg = (i % (G::HID / 8)) * 8;
    *(bf16x8*)&L.W2Tt[c][jg] = *(const bf16x8*)&w2t[c * G::HID + jg];
  }
}

template <typename G>
__device__ __forceinline__ void zero_dl_pad(const G& L) {
  if constexpr (G::CPAD < 32) {
    constexpr int PADW = 32 - G::CPAD;
    for (int i = threadIdx.x; i < G::RT * PADW; i += G::BLOCK) {
      const int r = i / PADW, c = G::CPAD + (i % PADW);
      L.DLs[r][c] = 0;
    }
  }
}

// ---------------------------------------------------------------------------
// fused generalized step kernel
// ---------------------------------------------------------------------------

template <int HID, int RT, int CP>
__global__ void __launch_bounds__(512)
mlp_step_gen_kernel(const u16* __restrict__ Xbf,   // [N][inp] staged bf16 (zero-padded)
                    const int* __restrict__ y, int B,
                    int inp, int cls,
                    const u16* __restrict__ wimg,  // packed weight images
                    const float* __restrict__ master,  // biases prefetch
                    float* __restrict__ slabs, int slab_stride,
                    float invBtot) {
  using G = GG<HID, RT, CP>;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  G L;
  L.carve(smem);

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4, lr_ = l & 15;
  const long long row0 = (long long)blockIdx.x * RT;

  const int off_b1 = inp * HID;
  const int off_w2 = off_b1 + HID;
  const int off_b2 = off_w2 + HID * G::CPAD;
  const int nparam = off_b2 + G::CPAD;

  for (int i = tid; i < HID; i += G::BLOCK) L.db1[i] = 0.f;
  if (tid < G::CPAD) L.db2[tid] = 0.f;
  if (tid == 0) L.loss[0] = 0.f;
  zero_dl_pad(L);
  load_w2_images(L, wimg, inp);
  for (int i = tid; i < HID; i += G::BLOCK) L.b1s[i] = master[off_b1 + i];
  if (tid < G::CPAD) L.b2s[tid] = master[off_b2 + tid];
  __syncthreads();

  // ---- fwd1: H^T = W1T @ B(Xs), K streamed in TK tiles over inp.
  // Double-buffered where LDS allows (G::DB): the NEXT tile's
  // cooperative load issues before this tile's MFMAs, so the L2/HBM
  // round trip overlaps compute instead of serializing each k step.
  f32x4 acc1[G::SLOTS];
  #pragma unroll
  for (int s = 0; s < G::SLOTS; ++s) acc1[s] = (f32x4){0.f, 0.f, 0.f, 0.f};
  const int nkt = (inp + G::TK - 1) / G::TK;
  load_xs_tile(L, 0, Xbf, inp, row0, (long long)B, 0, min(G::TK, inp));
  load_w1t_tile(L, 0, wimg, inp, 0, min(G::TK, inp));
  __syncthreads();
  for (int kt = 0; kt < nkt; ++kt) {
    const int cur = G::DB ? (kt & 1) : 0;
    if (G::DB && kt + 1 < nkt) {
      const int k0n = (kt + 1) * G::TK;
      const int kvn = min(G::TK, inp - k0n);
      load_xs_tile(L, cur ^ 1, Xbf, inp, row0, (long long)B, k0n, kvn);
      load_w1t_tile(L, cur ^ 1, wimg, inp, k0n, kvn);
    }
    u16 (*W1Tc)[G::W1S] = L.W1T(cur);
    u16 (*Xsc)[G::XSS] = L.Xs(cur);
    #pragma unroll
    for (int s = 0; s < G::SLOTS; ++s) {
      const int t = wave + G::WAVES * s;
      if (t < G::NT1) {
        const int mt = t % (HID / 16);
        const int rt = t / (HID / 16);
        f32x4 acc = acc1[s];
        #pragma unroll
        for (int ks = 0; ks < G::TK / 32; ++ks) {
          const bf16x8 a = *(const bf16x8*)&W1Tc[mt * 16 + lr_][ks * 32 + lg * 8];
          const bf16x8 b = *(const bf16x8*)&Xsc[rt * 16 + lr_][ks * 32 + lg * 8];
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
        }
        acc1[s] = acc;
      }
    }
    __syncthreads();  // prefetched buffer complete / current reads done
    if (!G::DB && kt + 1 < nkt) {
      const int k0n = (kt + 1) * G::TK;
      const int kvn = min(G::TK, inp - k0n);
      load_xs_tile(L, 0, Xbf, inp, row0, (long long)B, k0n, kvn);
      load_w1t_tile(L, 0, wimg, inp, k0n, kvn);
      __syncthreads();
    }
  }
  // epilogue: bias + relu, store Hs and HT
  #pragma unroll
  for (int s = 0; s < G::SLOTS; ++s) {
    const int t = wave + G::WAVES * s;
    if (t < G::NT1) {
      const int mt = t % (HID / 16);
      const int rt = t / (HID / 16);
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int h = mt * 16 + lg * 4 + r;
        const int row = rt * 16 + lr_;
        float hv = acc1[s][r] + L.b1s[h];
        hv = hv > 0.f ? hv : 0.f;
        const u16 hb = f2bf(hv);
        L.Hs[row][h] = hb;
        L.HT[h][row] = hb;
      }
    }
  }
  __syncthreads();

  // prefetch dW1's FIRST transposed-X tile now: xbuf is free (fwd1's
  // last Xs read retired at the barrier above) and nothing between
  // here and dW1 touches it, so the load's round trip hides under the
  // fwd2/dH/dW2 phases
  load_xt_tile(L, 0, Xbf, inp, row0, (long long)B, 0, min(G::TK, inp));

  // ---- fwd2: L^T = W2T @ B(Hs); wave-shuffle softmax + xent over NCT
  // class tiles (classes <= 16: one MFMA tile; 17..32: two tiles, the
  // max/sum folds across the register-resident tile axis first, then
  // the 4 lane-groups) ------------------------------------------------------
  for (int t2 = wave; t2 < RT / 16; t2 += G::WAVES) {
    f32x4 acc[G::NCT];
    #pragma unroll
    for (int ct = 0; ct < G::NCT; ++ct) {
      acc[ct] = (f32x4){0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int ks = 0; ks < HID / 32; ++ks) {
        const bf16x8 a = *(const bf16x8*)&L.W2Tt[ct * 16 + lr_][ks * 32 + lg * 8];
        const bf16x8 b = *(const bf16x8*)&L.Hs[t2 * 16 + lr_][ks * 32 + lg * 8];
        acc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[ct], 0, 0, 0);
      }
    }
    const int row = t2 * 16 + lr_;
    const bool valid = (row0 + row) < (long long)B;
    const int label = valid ? y[row0 + row] : -1;

    float logit[G::NCT][4];
    float mx = -1e30f;
    #pragma unroll
    for (int ct = 0; ct < G::NCT; ++ct) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int c = ct * 16 + lg * 4 + r;
        logit[ct][r] = (c < cls) ? acc[ct][r] + L.b2s[c] : -1e30f;
        mx = fmaxf(mx, logit[ct][r]);
      }
    }
    mx = fmaxf(mx, __shfl_xor(mx, 16, 64));
    mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
    float e[G::NCT][4], ssum = 0.f;
    #pragma unroll
    for (int ct = 0; ct < G::NCT; ++ct) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int c = ct * 16 + lg * 4 + r;
        e[ct][r] = (c < cls) ? __expf(logit[ct][r] - mx) : 0.f;
        ssum += e[ct][r];
      }
    }
    ssum += __shfl_xor(ssum, 16, 64);
    ssum += __shfl_xor(ssum, 32, 64);
    const float rs = fast_rcp(ssum);
    const float logs = __logf(ssum);
    float db2_acc[G::NCT][4];
    float loss_acc = 0.f;
    #pragma unroll
    for (int ct = 0; ct < G::NCT; ++ct) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int c = ct * 16 + lg * 4 + r;
        const float dl =
            valid ? (e[ct][r] * rs - (c == label ? 1.f : 0.f)) * invBtot : 0.f;
        const u16 dlb = f2bf(dl);
        L.DLs[row][c] = dlb;
        L.DLT[c][row] = dlb;
        db2_acc[ct][r] = dl;
        if (valid && c == label) loss_acc = -(logit[ct][r] - mx - logs) * invBtot;
      }
    }
    #pragma unroll
    for (int bit = 1; bit < 16; bit <<= 1) {
      #pragma unroll
      for (int ct = 0; ct < G::NCT; ++ct) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) db2_acc[ct][r] += __shfl_xor(db2_acc[ct][r], bit, 64);
      }
      loss_acc += __shfl_xor(loss_acc, bit, 64);
    }
    if (lr_ == 0) {
      #pragma unroll
      for (int ct = 0; ct < G::NCT; ++ct) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          atomicAdd(&L.db2[ct * 16 + lg * 4 + r], db2_acc[ct][r]);
        }
      }
    }
    loss_acc += __shfl_xor(loss_acc, 16, 64);
    loss_acc += __shfl_xor(loss_acc, 32, 64);
    if (l == 0) atomicAdd(L.loss, loss_acc);
  }
  __syncthreads();

  // ---- dH^T = W2s @ B(DLs) + relu mask + db1 -------------------------------
  #pragma unroll
  for (int s = 0; s < G::SLOTS; ++s) {
    const int t = wave + G::WAVES * s;
    if (t < G::NT1) {
      const int mt = t % (HID / 16);
      const int rt = t / (HID / 16);
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      const bf16x8 a = *(const bf16x8*)&L.W2s[mt * 16 + lr_][lg * 8];
      const bf16x8 b = *(const bf16x8*)&L.DLs[rt * 16 + lr_][lg * 8];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int h = mt * 16 + lg * 4 + r;
        const int row = rt * 16 + lr_;
        const float hv = bf2f(L.HT[h][row]);
        float dh = hv > 0.f ? acc[r] : 0.f;
        L.DHT[h][row] = f2bf(dh);
        dh += __shfl_xor(dh, 1, 64);
        dh += __shfl_xor(dh, 2, 64);
        dh += __shfl_xor(dh, 4, 64);
        dh += __shfl_xor(dh, 8, 64);
        if (lr_ == 0) atomicAdd(&L.db1[h], dh);
      }
    }
  }
  __syncthreads();  // DHT complete

  float* slab = slabs + (long long)blockIdx.x * slab_stride;

  // ---- dW2 = HT @ B(DLT) -> slab (plain stores) ----------------------------
  for (int t = wave; t < (HID / 16) * G::NCT; t += G::WAVES) {
    const int ht = t % (HID / 16);
    const int ct = t / (HID / 16);
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int ks = 0; ks < RT / 32; ++ks) {
      const bf16x8 a = *(const bf16x8*)&L.HT[ht * 16 + lr_][ks * 32 + lg * 8];
      const bf16x8 b = *(const bf16x8*)&L.DLT[ct * 16 + lr_][ks * 32 + lg * 8];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int h = ht * 16 + lg * 4 + r;
      slab[off_w2 + h * G::CPAD + ct * 16 + lr_] = acc[r];
    }
  }

  // ---- dW1^T = DHT @ B(XT), K'-streamed over inp -> slab -------------------
  // (tile 0 was prefetched before fwd2; the barrier below also covers
  // the dH-phase atomics)
  __syncthreads();
  for (int kt = 0; kt < nkt; ++kt) {
    const int cur = G::DB ? (kt & 1) : 0;
    if (G::DB && kt + 1 < nkt) {
      const int k0n = (kt + 1) * G::TK;
      load_xt_tile(L, cur ^ 1, Xbf, inp, row0, (long long)B, k0n,
                   min(G::TK, inp - k0n));
    }
    const int k0 = kt * G::TK;
    const int kv = min(G::TK, inp - k0);
    u16 (*XTc)[G::XTS] = L.XT(cur);
    const int ntw = (HID / 16) * (kv / 16);
    for (int t = wave; t < ntw; t += G::WAVES) {
      const int mt = t % (HID / 16);
      const int it = t / (HID / 16);
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int ks = 0; ks < RT / 32; ++ks) {
        const bf16x8 a = *(const bf16x8*)&L.DHT[mt * 16 + lr_][ks * 32 + lg * 8];
        const bf16x8 b = *(const bf16x8*)&XTc[it * 16 + lr_][ks * 32 + lg * 8];
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
      }
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int h = mt * 16 + lg * 4 + r;
        const int in = k0 + it * 16 + lr_;
        slab[in * HID + h] = acc[r];
      }
    }
    __syncthreads();
    if (!G::DB && kt + 1 < nkt) {
      const int k0n = (kt + 1) * G::TK;
      load_xt_tile(L, 0, Xbf, inp, row0, (long long)B, k0n,
                   min(G::TK, inp - k0n));
      __syncthreads();
    }
  }

  // ---- bias grads + loss into the slab -------------------------------------
  for (int i = tid; i < HID; i += G::BLOCK) slab[off_b1 + i] = L.db1[i];
  if (tid < G::CPAD) slab[off_b2 + tid] = L.db2[tid];
  if (tid == G::CPAD) slab[nparam] = L.loss[0];

}

// ---------------------------------------------------------------------------
// wide-grid reduce + Adam: sums the per-WG slabs and applies Adam (or,
// in grads_out mode, hands the summed grads to the RCCL all-reduce).
//
// Design note vs the specialized kernel's single-launch handshake
// (tabular_kernels.hip G16): at generalized geometries nparam is large
// (MNIST shape: ~102k params -> ~8 MB of slab+state traffic per step)
// and the step grid is tiny (ceil(B/RT) WGs), so an in-kernel reduce is
// bandwidth-starved — memory parallelism scales with resident CUs. A
// separate kernel launches with hundreds of WGs, reaches full-chip
// bandwidth, and the kernel boundary IS the inter-WG barrier, so the
// epoch-tag handshake disappears. Stream-ordered + hipGraph-capturable;
// t advances via bump_t_kernel so every WG sees one consistent t.
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256)
reduce_adam_gen_kernel(const float* __restrict__ slabs, int n_wg,
                       int slab_stride, int nparam, int inp, int hid, int cpad,
                       float* __restrict__ master, u16* __restrict__ bfmirror,
                       float* __restrict__ m, float* __restrict__ v,
                       int* __restrict__ t_dev,
                       float* __restrict__ loss_out,
                       float lr, float beta1, float beta2, float eps,
                       u16* __restrict__ wimg,
                       float* __restrict__ grads_out,
                       unsigned* __restrict__ done_counter) {
  const float t_new = (float)(*t_dev + 1);
  const float corr1 = fast_rcp(1.f - __powf(beta1, t_new));
  const float corr2 = fast_rcp(1.f - __powf(beta2, t_new));
  const int off_b1 = inp * hid;
  const int off_w2 = off_b1 + hid;
  const int off_b2 = off_w2 + hid * cpad;
  for (int i = blockIdx.x * 256 + threadIdx.x; i <= nparam;
       i += gridDim.x * 256) {
    float g0 = 0.f, g1 = 0.f, g2 = 0.f, g3 = 0.f;
    int w = 0;
    for (; w + 4 <= n_wg; w += 4) {
      g0 += slabs[(long long)w * slab_stride + i];
      g1 += slabs[(long long)(w + 1) * slab_stride + i];
      g2 += slabs[(long long)(w + 2) * slab_stride + i];
      g3 += slabs[(long long)(w + 3) * slab_stride + i];
    }
    for (; w < n_wg; ++w) g0 += slabs[(long long)w * slab_stride + i];
    const float g = (g0 + g1) + (g2 + g3);
    if (grads_out) {          // reduce-only: RCCL takes it from here
      grads_out[i] = g;
      continue;
    }
    if (i == nparam) {
      *loss_out = g;
      continue;
    }
    const float mi = beta1 * m[i] + (1.f - beta1) * g;
    const float vi = beta2 * v[i] + (1.f - beta2) * g * g;
    m[i] = mi;
    v[i] = vi;
    const float p = master[i] - lr * (mi * corr1) * fast_rcp(sqrtf(vi * corr2) + eps);
    master[i] = p;
    const u16 wb = f2bf(p);
    bfmirror[i] = wb;
    if (wimg) {
      if (i < off_b1) {
        wimg[(i % hid) * inp + (i / hid)] = wb;
      } else if (i >= off_w2 && i < off_b2) {
        const int j = i - off_w2;
        const int h = j / cpad, c = j % cpad;
        wimg[hid * inp + h * 32 + c] = wb;
        wimg[hid * inp + hid * 32 + c * hid + h] = wb;
      }
    }
  }
  // advance the Adam step counter in-kernel: every WG read t_dev at
  // entry (the last-done WG writes only after all WGs passed their
  // read), saving the separate 1-thread bump kernel (~4.7 us/launch
  // of pure launch overhead per step)
  if (threadIdx.x == 0) {
    const unsigned done = atomicAdd(done_counter, 1u) + 1u;
    if (done == gridDim.x) {
      *done_counter = 0u;
      if (!grads_out) *t_dev = (int)t_new;
    }
  }
}

// ---------------------------------------------------------------------------
// generalized fused predict: standardize + fwd + argmax (+softmax)
// ---------------------------------------------------------------------------

template <int HID, int RT, int CP>
__global__ void __launch_bounds__(512)
mlp_predict_gen_kernel(const float* __restrict__ X,  // [B][draw] raw fp32
                       int B, int draw, int inp, int cls,
                       const float* __restrict__ mean,
                       const float* __restrict__ invstd,
                       const u16* __restrict__ wimg,
                       const float* __restrict__ master,  // biases
                       int* __restrict__ preds,
                       float* __restrict__ probs /* optional [B][cls] */) {
  using G = GG<HID, RT, CP>;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  G L;
  L.carve(smem);

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4, lr = l & 15;
  const long long row0 = (long long)blockIdx.x * RT;

  const int off_b1 = inp * HID;
  const int off_b2 = off_b1 + HID + HID * G::CPAD;

  load_w2_images(L, wimg, inp);
  for (int i = tid; i < HID; i += G::BLOCK) L.b1s[i] = master[off_b1 + i];
  if (tid < G::CPAD) L.b2s[tid] = master[off_b2 + tid];
  __syncthreads();

  // fwd1 with standardize-on-load, K streamed
  f32x4 acc1[G::SLOTS];
  #pragma unroll
  for (int s = 0; s < G::SLOTS; ++s) acc1[s] = (f32x4){0.f, 0.f, 0.f, 0.f};
  const int nkt = (inp + G::TK - 1) / G::TK;
  u16 (*Xs0)[G::XSS] = L.Xs(0);
  u16 (*W1T0)[G::W1S] = L.W1T(0);
  for (int kt = 0; kt < nkt; ++kt) {
    const int k0 = kt * G::TK;
    for (int i = tid; i < RT * G::TK; i += G::BLOCK) {
      const int r = i / G::TK, j = i % G::TK;
      const int c = k0 + j;
      float v = 0.f;
      if (row0 + r < B && c < draw) {
        v = (X[(row0 + r) * draw + c] - mean[c]) * invstd[c];
      }
      Xs0[r][j] = f2bf(v);
    }
    load_w1t_tile(L, 0, wimg, inp, k0, min(G::TK, inp - k0));
    __syncthreads();
    #pragma unroll
    for (int s = 0; s < G::SLOTS; ++s) {
      const int t = wave + G::WAVES * s;
      if (t < G::NT1) {
        const int mt = t % (HID / 16);
        const int rt = t / (HID / 16);
        f32x4 acc = acc1[s];
        #pragma unroll
        for (int ks = 0; ks < G::TK / 32; ++ks) {
          const bf16x8 a = *(const bf16x8*)&W1T0[mt * 16 + lr][ks * 32 + lg * 8];
          const bf16x8 b = *(const bf16x8*)&Xs0[rt * 16 + lr][ks * 32 + lg * 8];
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
        }
        acc1[s] = acc;
      }
    }
    __syncthreads();
  }
  #pragma unroll
  for (int s = 0; s < G::SLOTS; ++s) {
    const int t = wave + G::WAVES * s;
    if (t < G::NT1) {
      const int mt = t % (HID / 16);
      const int rt = t / (HID / 16);
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int h = mt * 16 + lg * 4 + r;
        const int row = rt * 16 + lr;
        float hv = acc1[s][r] + L.b1s[h];
        L.Hs[row][h] = f2bf(hv > 0.f ? hv : 0.f);
      }
    }
  }
  __syncthreads();

  // fwd2 + argmax / softmax over NCT class tiles
  for (int t2 = wave; t2 < RT / 16; t2 += G::WAVES) {
    f32x4 acc[G::NCT];
    #pragma unroll
    for (int ct = 0; ct < G::NCT; ++ct) {
      acc[ct] = (f32x4){0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int ks = 0; ks < HID / 32; ++ks) {
        const bf16x8 a = *(const bf16x8*)&L.W2Tt[ct * 16 + lr][ks * 32 + lg * 8];
        const bf16x8 b = *(const bf16x8*)&L.Hs[t2 * 16 + lr][ks * 32 + lg * 8];
        acc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[ct], 0, 0, 0);
      }
    }
    const int row = t2 * 16 + lr;
    float best = -1e30f;
    int bcol = cls;
    float logit[G::NCT][4];
    #pragma unroll
    for (int ct = 0; ct < G::NCT; ++ct) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int c = ct * 16 + lg * 4 + r;
        logit[ct][r] = (c < cls) ? acc[ct][r] + L.b2s[c] : -1e30f;
        // ties pick the lowest class (match torch.argmax)
        if (logit[ct][r] > best) { best = logit[ct][r]; bcol = c; }
      }
    }
    #pragma unroll
    for (int d = 16; d < 64; d <<= 1) {
      const float ov = __shfl_xor(best, d, 64);
      const int oc = __shfl_xor(bcol, d, 64);
      if (ov > best || (ov == best && oc < bcol)) { best = ov; bcol = oc; }
    }
    if (lg == 0 && row0 + row < B) preds[row0 + row] = bcol;
    if (probs != nullptr) {
      float ssum = 0.f, e[G::NCT][4];
      #pragma unroll
      for (int ct = 0; ct < G::NCT; ++ct) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int c = ct * 16 + lg * 4 + r;
          e[ct][r] = (c < cls) ? __expf(logit[ct][r] - best) : 0.f;
          ssum += e[ct][r];
        }
      }
      ssum += __shfl_xor(ssum, 16, 64);
      ssum += __shfl_xor(ssum, 32, 64);
      const float rs = fast_rcp(ssum);
      #pragma unroll
      for (int ct = 0; ct < G::NCT; ++ct) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int c = ct * 16 + lg * 4 + r;
          if (c < cls && row0 + row < B) probs[(row0 + row) * cls + c] = e[ct][r] * rs;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// generalized Adam (DP path): grid-strided over a runtime param count.
// The step counter is advanced by a separate 1-thread kernel launched
// AFTER this one (stream-ordered) so every WG sees the same t.
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256)
adam_step_gen_kernel(float* __restrict__ master, u16* __restrict__ bfmirror,
                     const float* __restrict__ grads, float* __restrict__ m,
                     float* __restrict__ v, const int* __restrict__ t_dev,
                     int nparam, int inp, int hid, int cpad,
                     float lr, float beta1, float beta2, float eps,
                     u16* __restrict__ wimg) {
  const float t = (float)(*t_dev + 1);
  const float corr1 = fast_rcp(1.f - __powf(beta1, t));
  const float corr2 = fast_rcp(1.f - __powf(beta2, t));
  const int off_b1 = inp * hid;
  const int off_w2 = off_b1 + hid;
  const int off_b2 = off_w2 + hid * cpad;
  for (int i = blockIdx.x * 256 + threadIdx.x; i < nparam; i += gridDim.x * 256) {
    const float g = grads[i];
    const float mi = beta1 * m[i] + (1.f - beta1) * g;
    const float vi = beta2 * v[i] + (1.f - beta2) * g * g;
    m[i] = mi;
    v[i] = vi;
    const float p = master[i] - lr * (mi * corr1) * fast_rcp(sqrtf(vi * corr2) + eps);
    master[i] = p;
    const u16 wb = f2bf(p);
    bfmirror[i] = wb;
    if (wimg) {
      if (i < off_b1) {
        wimg[(i % hid) * inp + (i / hid)] = wb;
      } else if (i >= off_w2 && i < off_b2) {
        const int j = i - off_w2;
        const int h = j / cpad, c = j % cpad;
        wimg[hid * inp + h * 32 + c] = wb;
        wimg[hid * inp + hid * 32 + c * hid + h] = wb;
      }
    }
  }
}

__global__ void bump_t_kernel(int* __restrict__ t_dev) { ++(*t_dev); }

}  // namespace gen

// ---------------------------------------------------------------------------
// host launchers (extern "C"; stream-ordered, capture-safe)
// ---------------------------------------------------------------------------

namespace {

int rt_for_hid(int hid) {
  switch (hid) {
    case 32: return 128;
    case 64: return 128;
    case 128: return 64;
    case 256: return 32;
  }
  return 0;
}

template <int HID, int RT, int CP>
int launch_step_gen_t(const unsigned short* Xbf, const int* y, int B, int inp,
                      int cls, const unsigned short* wimg, const float* master,
                      float* slabs, int slab_stride, int max_slabs,
                      float invBtot, hipStream_t stream) {
  using G = gen::GG<HID, RT, CP>;
  const int blocks = (B + RT - 1) / RT;
  if (blocks > max_slabs) return -1;
  static int done = 0;
  if (!done) {
    if (hipFuncSetAttribute((const void*)gen::mlp_step_gen_kernel<HID, RT, CP>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            G::TOTAL) != hipSuccess) {
      return -2;
    }
    done = 1;
  }
  hipLaunchKernelGGL((gen::mlp_step_gen_kernel<HID, RT, CP>), dim3(blocks),
                     dim3(G::BLOCK), G::TOTAL, stream, Xbf, y, B, inp, cls, wimg,
                     master, slabs, slab_stride, invBtot);
  return 0;
}

template <int HID, int RT, int CP>
int launch_predict_gen_t(const float* X, int B, int draw, int inp, int cls,
                         const float* mean, const float* invstd,
                         const unsigned short* wimg, const float* master,
                         int* preds, float* probs, hipStream_t stream) {
  using G = gen::GG<HID, RT, CP>;
  static int done = 0;
  if (!done) {
    if (hipFuncSetAttribute(
            (const void*)gen::mlp_predict_gen_kernel<HID, RT, CP>,
            hipFuncAttributeMaxDynamicSharedMemorySize, G::TOTAL) != hipSuccess) {
      return -2;
    }
    done = 1;
  }
  const int blocks = (B + RT - 1) / RT;
  hipLaunchKernelGGL((gen::mlp_predict_gen_kernel<HID, RT, CP>), dim3(blocks),
                     dim3(G::BLOCK), G::TOTAL, stream, X, B, draw, inp, cls, mean,
                     invstd, wimg, master, preds, probs);
  return 0;
}

}  // namespace

extern "C" {

// supported hidden widths (template instantiations); returns rows/WG or 0
int gen_rt_for_hid(int hid) { return rt_for_hid(hid); }

int launch_mlp_step_gen(const unsigned short* Xbf, const int* y, int B, int inp,
                        int hid, int cls, int rt, const unsigned short* wimg,
                        const float* master, float* slabs, int slab_stride,
                        int max_slabs, float invBtot, hipStream_t stream) {
  if (inp % 32 != 0 || cls > 32) return -3;
  #define STEP_CASE(H, R, C)                                                  \
    if (hid == H && rt == R && cls <= C && (C == 16 ? cls <= 16 : cls > 16))  \
      return launch_step_gen_t<H, R, C>(Xbf, y, B, inp, cls, wimg, master,    \
                                        slabs, slab_stride, max_slabs,        \
                                        invBtot, stream);
  STEP_CASE(32, 128, 16)
  STEP_CASE(32, 64, 16)
  STEP_CASE(64, 128, 16)
  STEP_CASE(64, 64, 16)
  STEP_CASE(128, 64, 16)
  STEP_CASE(128, 32, 16)
  STEP_CASE(256, 32, 16)
  STEP_CASE(32, 128, 32)
  STEP_CASE(64, 128, 32)
  STEP_CASE(128, 64, 32)
  STEP_CASE(256, 32, 32)
  #undef STEP_CASE
  return -3;
}

void launch_reduce_adam_gen(const float* slabs, int n_wg, int slab_stride,
                            int inp, int hid, int cpad, float* master,
                            unsigned short* bfmirror, float* m, float* v,
                            int* t_dev, float* loss_out, float lr, float beta1,
                            float beta2, float eps, unsigned short* wimg,
                            float* grads_out, unsigned* done_counter,
                            hipStream_t stream) {
  const int nparam = inp * hid + hid + hid * cpad + cpad;
  int blocks = (nparam + 255) / 256;
  if (blocks > 512) blocks = 512;
  hipLaunchKernelGGL(gen::reduce_adam_gen_kernel, dim3(blocks), dim3(256), 0,
                     stream, slabs, n_wg, slab_stride, nparam, inp, hid, cpad,
                     master, bfmirror, m, v, t_dev, loss_out, lr, beta1, beta2,
                     eps, wimg, grads_out, done_counter);
}

int launch_mlp_predict_gen(const float* X, int B, int draw, int inp, int hid,
                           int cls, const float* mean, const float* invstd,
                           const unsigned short* wimg, const float* master,
                           int* preds, float* probs, hipStream_t stream) {
  if (inp % 32 != 0 || cls > 32) return -3;
  #define PRED_CASE(H, R, C)                                                  \
    if (hid == H && (C == 16 ? cls <= 16 : cls > 16))                         \
      return launch_predict_gen_t<H, R, C>(X, B, draw, inp, cls, mean, invstd,\
                                           wimg, master, preds, probs, stream);
  PRED_CASE(32, 128, 16)
  PRED_CASE(64, 128, 16)
  PRED_CASE(128, 64, 16)
  PRED_CASE(256, 32, 16)
  PRED_CASE(32, 128, 32)
  PRED_CASE(64, 128, 32)
  PRED_CASE(128, 64, 32)
  PRED_CASE(256, 32, 32)
  #undef PRED_CASE
  return -3;
}

void launch_adam_step_gen(float* master, unsigned short* bfmirror,
                          const float* grads, float* m, float* v, int* t_dev,
                          int nparam, int inp, int hid, int cpad, float lr,
                          float beta1, float beta2, float eps,
                          unsigned short* wimg, hipStream_t stream) {
  int blocks = (nparam + 255) / 256;
  if (blocks > 512) blocks = 512;
  hipLaunchKernelGGL(gen::adam_step_gen_kernel, dim3(blocks), dim3(256), 0,
                     stream, master, bfmirror, grads, m, v, t_dev, nparam, inp,
                     hid, cpad, lr, beta1, beta2, eps, wimg);
  hipLaunchKernelGGL(gen::bump_t_kernel, dim3(1), dim3(1), 0, stream, t_dev);
}

}  // extern "C"
