// Hand-written CDNA4 (gfx950) implicit-GEMM direct convolutions for the
// Atari conv stacks — NHWC, bf16 (optionally fused uint8
// dequant on the first layer), fused bias + ReLU epilogue.
//
// Replaces MIOpen igemm kernels for the fixed geometries of cfg/ape_x.json
// and cfg/impala.json (plus their R2D2 twin). Design:
//   * GEMM view: out[M=N*P*Q rows][COUT cols] = im2col(in)[M][K] x W[K][COUT],
//     K = KH*KW*C. MFMA v_mfma_f32_16x16x32_bf16, one wave computes a
//     16-row x COUT-col strip; a block = 4 waves = 64 rows.
//   * Weights (K x COUT, stored as W[cout][k] row-major = torch NHWC
//     weight.permute) are staged once into LDS as [cout][K+PAD] so B
//     fragments are ds_read_b128 with a conflict-breaking pad.
//   * A (im2col rows) is read straight from global: in NHWC every 32-wide
//     K-chunk of a patch row is 64 contiguous bytes (requires
//     (KW*C) % 32 == 0, true for all five shapes) -> one 16 B load per
//     lane per chunk; overlapping patches hit L2.
//   * Epilogue: bias + ReLU in f32, pack to bf16, scalar stores (outputs
//     are small).
//
// MFMA fragment maps (v_mfma_f32_16x16x32_bf16, guide §3):
//   A: lane l holds A[row=l&15][k=(l>>4)*8 + j], j=0..7
//   B: lane l holds B[k=(l>>4)*8 + j][col=l&15]
//   C/D: lane l holds C[row=(l>>4)*4 + r][col=l&15], r=0..3

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int kPad = 32;  // LDS row pad: (K+pad)*2 % 256 in {64,192} spreads the
// b128 16-lane service quads across all 16 bank-row slots (PMC-verified)

template <int H, int W, int C, int KH, int KW, int S, int COUT, bool U8IN>
struct ConvGeom {
  static constexpr int P = (H - KH) / S + 1;
  static constexpr int Q = (W - KW) / S + 1;
  static constexpr int K = KH * KW * C;
  static constexpr int KCHUNKS = K / 32;
  static_assert(K % 32 == 0, "K must be a multiple of 32");
  static_assert((KW * C) % 32 == 0, "patch row must tile by 32");
  static constexpr int ROWC = KW * C;          // elements per (kh) patch row
  static constexpr int LDS_ROW = K + kPad;     // per-cout LDS stride
  static constexpr int LDS_ELEMS = COUT * LDS_ROW;
};

// A block = WAVES waves, each computing RPW rows x COUT cols
// (RPW/16 row-fragments; B fragments shared across row-fragments).
// CHW_OUT: write the output pre-flattened in logical-NCHW element order
// ((n, c*P*Q + p*Q + q)) so the trailing torch.flatten of the conv stack
// becomes a free view instead of a channels_last->NCHW copy (measured
// 8.6 us x 3 forwards per Ape-X step).
template <int H, int W, int C, int KH, int KW, int S, int COUT, bool U8IN,
          int WAVES, int RPW, bool CHW_OUT = false>
__global__ __launch_bounds__(WAVES * 64) void conv_fwd_kernel(
    const void* __restrict__ in_v,       // (N,H,W,C) u8 or bf16 (NHWC)
    const __bf16* __restrict__ weight,   // (COUT, K) row-major NHWC flat
    const __bf16* __restrict__ bias,     // (COUT), may be null
    __bf16* __restrict__ out,            // (N,P,Q,COUT)
    int batch) {
  using G = ConvGeom<H, W, C, KH, KW, S, COUT, U8IN>;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* wlds = reinterpret_cast<__bf16*>(smem);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // ---- stage weights into LDS: [cout][K + pad] ----
  constexpr int WELEMS = COUT * G::K;
  for (int base = tid * 8; base < WELEMS; base += WAVES * 64 * 8) {
    int co = base / G::K;
    int k = base - co * G::K;
    bf16x8 v = *reinterpret_cast<const bf16x8*>(weight + co * G::K + k);
    *reinterpret_cast<bf16x8*>(wlds + co * G::LDS_ROW + k) = v;
  }
  __syncthreads();

  const int M = batch * G::P * G::Q;
  const int row0 = (blockIdx.x * WAVES + wave) * RPW;
  if (row0 >= M) return;
  constexpr int RFRAG = RPW / 16;
  constexpr int NFRAG = COUT / 16;
  // per-lane A row bases, one per row-fragment (clamped for ragged tail)
  int64_t in_row0[RFRAG];
#pragma unroll
  for (int rf = 0; rf < RFRAG; ++rf) {
    int arow = row0 + rf * 16 + (lane & 15);
    if (arow >= M) arow = M - 1;
    const int n = arow / (G::P * G::Q);
    const int rem = arow - n * (G::P * G::Q);
    const int p = rem / G::Q;
    const int q = rem - p * G::Q;
    in_row0[rf] = ((int64_t)n * H + p * S) * (W * C) + q * S * C;
  }
  const int kpart = (lane >> 4) * 8;

  f32x4 acc[RFRAG][NFRAG];
#pragma unroll
  for (int rf = 0; rf < RFRAG; ++rf)
#pragma unroll
    for (int f = 0; f < NFRAG; ++f) acc[rf][f] = {0.f, 0.f, 0.f, 0.f};

#pragma unroll
  for (int kc = 0; kc < G::KCHUNKS; ++kc) {
    const int kelem = kc * 32 + kpart;
    const int dy = kelem / G::ROWC;
    const int dx = kelem - dy * G::ROWC;
    const int64_t koff = (int64_t)dy * (W * C) + dx;
    bf16x8 a[RFRAG];
#pragma unroll
    for (int rf = 0; rf < RFRAG; ++rf) {
      const int64_t goff = in_row0[rf] + koff;
      if constexpr (U8IN) {
        const uint8_t* src = reinterpret_cast<const uint8_t*>(in_v) + goff;
        uint2 raw = *reinterpret_cast<const uint2*>(src);
        const float inv255 = 1.0f / 255.0f;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          unsigned byte =
              (j < 4 ? raw.x >> (8 * j) : raw.y >> (8 * (j - 4))) & 0xFF;
          a[rf][j] = (__bf16)(byte * inv255);
        }
      } else {
        a[rf] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const __bf16*>(in_v) + goff);
      }
    }
#pragma unroll
    for (int f = 0; f < NFRAG; ++f) {
      const int col = f * 16 + (lane & 15);
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          wlds + col * G::LDS_ROW + kc * 32 + kpart);
#pragma unroll
      for (int rf = 0; rf < RFRAG; ++rf)
        acc[rf][f] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[rf], b, acc[rf][f], 0, 0, 0);
    }
  }

  // ---- epilogue: bias + ReLU + bf16 store ----
  const int crow_base = (lane >> 4) * 4;
#pragma unroll
  for (int rf = 0; rf < RFRAG; ++rf)
#pragma unroll
    for (int f = 0; f < NFRAG; ++f) {
      const int col = f * 16 + (lane & 15);
      const float bv = bias ? (float)bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int orow = row0 + rf * 16 + crow_base + r;
        if (orow < M) {
          float v = acc[rf][f][r] + bv;
          v = v > 0.0f ? v : 0.0f;
          if constexpr (CHW_OUT) {
            const int n = orow / (G::P * G::Q);
            const int pq = orow - n * (G::P * G::Q);
            out[((int64_t)n * COUT + col) * (G::P * G::Q) + pq] = (__bf16)v;
          } else {
            out[(int64_t)orow * COUT + col] = (__bf16)v;
          }
        }
      }
    }
}

struct ConvLaunch {
  int H, W, C, KH, KW, S, COUT;
  bool u8;
  void (*fn)(const void*, const __bf16*, const __bf16*, __bf16*, int);
  int lds_bytes;
  int waves, rpw;
};

template <int H, int W, int C, int KH, int KW, int S, int COUT, bool U8,
          int WAVES = 4, int RPW = 16, bool CHW = false>
ConvLaunch make_launch() {
  using G = ConvGeom<H, W, C, KH, KW, S, COUT, U8>;
  return ConvLaunch{H, W, C, KH, KW, S, COUT, U8,
                    conv_fwd_kernel<H, W, C, KH, KW, S, COUT, U8, WAVES, RPW,
                                    CHW>,
                    (int)(G::LDS_ELEMS * sizeof(__bf16)), WAVES, RPW};
}

// CHW-out variants for the stack-final conv of each model (flatten fusion)
static const ConvLaunch kChwLaunches[] = {
    make_launch<9, 9, 64, 3, 3, 1, 64, false, 8, 16, true>(),
    make_launch<20, 20, 16, 4, 4, 2, 32, false, 8, 16, true>(),
};

static const ConvLaunch kLaunches[] = {
    // Ape-X / R2D2 stack (cfg/ape_x.json:38-51); WAVES/RPW picked by the
    // variant sweep in tools/gpu_conv_tune.py (profiles/)
    make_launch<84, 84, 4, 8, 8, 4, 32, true, 8, 32>(),
    make_launch<84, 84, 4, 8, 8, 4, 32, false, 8, 32>(),
    make_launch<20, 20, 32, 4, 4, 2, 64, false, 8, 16>(),
    make_launch<9, 9, 64, 3, 3, 1, 64, false, 8, 16>(),
    // IMPALA stack (cfg/impala.json:26-39)
    make_launch<84, 84, 4, 8, 8, 4, 16, true, 8, 16>(),
    make_launch<84, 84, 4, 8, 8, 4, 16, false, 8, 16>(),
    make_launch<20, 20, 16, 4, 4, 2, 32, false, 8, 16>(),
};

// variant table for on-GPU tuning (tools/gpu_conv_tune.py)
static const ConvLaunch kVariants[] = {
    make_launch<84, 84, 4, 8, 8, 4, 32, true, 4, 16>(),
    make_launch<84, 84, 4, 8, 8, 4, 32, true, 8, 16>(),
    make_launch<84, 84, 4, 8, 8, 4, 32, true, 4, 32>(),
    make_launch<84, 84, 4, 8, 8, 4, 32, true, 8, 32>(),
    make_launch<20, 20, 32, 4, 4, 2, 64, false, 4, 16>(),
    make_launch<20, 20, 32, 4, 4, 2, 64, false, 8, 16>(),
    make_launch<20, 20, 32, 4, 4, 2, 64, false, 4, 32>(),
    make_launch<20, 20, 32, 4, 4, 2, 64, false, 8, 32>(),
    make_launch<9, 9, 64, 3, 3, 1, 64, false, 4, 16>(),
    make_launch<9, 9, 64, 3, 3, 1, 64, false, 8, 16>(),
    make_launch<9, 9, 64, 3, 3, 1, 64, false, 4, 32>(),
    make_launch<9, 9, 64, 3, 3, 1, 64, false, 8, 32>(),
};

// Layout probe: C(16,16) = A(16,32) x B(32,16) with exactly the fragment
// maps documented above — unit-tested against torch.matmul on the GPU so a
// map error fails loudly instead of silently transposing outputs.
__global__ void mfma_probe_kernel(const __bf16* __restrict__ A,
                                  const __bf16* __restrict__ B,
                                  float* __restrict__ C) {
  int lane = threadIdx.x & 63;
  bf16x8 a, b;
  int kpart = (lane >> 4) * 8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(lane & 15) * 32 + kpart + j];   // A[row][k]
    b[j] = B[(kpart + j) * 16 + (lane & 15)]; // B[k][col]
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    C[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

}  // namespace

void mfma_probe(torch::Tensor A, torch::Tensor B, torch::Tensor C) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)at::cuda::getCurrentCUDAStream().stream(),
                     (const __bf16*)A.data_ptr(), (const __bf16*)B.data_ptr(),
                     C.data_ptr<float>());
}

// Returns true if a fused kernel exists for this geometry.
bool conv_fwd_supported(int64_t H, int64_t W, int64_t C, int64_t KH, int64_t KW,
                        int64_t S, int64_t COUT, bool u8) {
  for (const auto& l : kLaunches)
    if (l.H == H && l.W == W && l.C == C && l.KH == KH && l.KW == KW &&
        l.S == S && l.COUT == COUT && l.u8 == u8)
      return true;
  return false;
}

// in: (N,C,H,W) logical, channels_last (or uint8 NCHW-contiguous when u8 —
// then treated as NHWC? no: u8 input must also be channels_last). weight:
// (COUT,C,KH,KW) logical channels_last. bias: (COUT) f32 or undefined.
// out: (N,COUT,P,Q) logical channels_last bf16.
void conv_fwd(torch::Tensor in, torch::Tensor weight, torch::Tensor bias,
              torch::Tensor out, int64_t stride) {
  TORCH_CHECK(in.is_cuda() && out.is_cuda());
  TORCH_CHECK(in.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv_fwd input must be channels_last");
  TORCH_CHECK(out.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(weight.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int N = (int)in.size(0), C = (int)in.size(1), H = (int)in.size(2),
            W = (int)in.size(3);
  const int COUT = (int)weight.size(0), KH = (int)weight.size(2),
            KW = (int)weight.size(3);
  const bool u8 = in.scalar_type() == torch::kUInt8;
  if (!u8) TORCH_CHECK(in.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(weight.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16);
  const ConvLaunch* L = nullptr;
  for (const auto& l : kLaunches)
    if (l.H == H && l.W == W && l.C == C && l.KH == KH && l.KW == KW &&
        l.S == (int)stride && l.COUT == COUT && l.u8 == u8) {
      L = &l;
      break;
    }
  TORCH_CHECK(L, "no fused conv kernel for this geometry: ", H, "x", W, "x", C,
              " k", KH, "x", KW, " s", stride, " -> ", COUT, " u8=", u8);
  const int P = (H - KH) / (int)stride + 1, Q = (W - KW) / (int)stride + 1;
  const int M = N * P * Q;
  const int rows_per_block = L->waves * L->rpw;
  const int blocks = (M + rows_per_block - 1) / rows_per_block;
  const __bf16* bias_ptr = nullptr;
  if (bias.defined() && bias.numel() > 0) {
    TORCH_CHECK(bias.scalar_type() == torch::kBFloat16);
    bias_ptr = (const __bf16*)bias.data_ptr();
  }
  hipLaunchKernelGGL(L->fn, dim3(blocks), dim3(L->waves * 64), L->lds_bytes,
                     (hipStream_t)at::cuda::getCurrentCUDAStream().stream(),
                     (const void*)in.data_ptr(),
                     (const __bf16*)weight.data_ptr(), bias_ptr,
                     (__bf16*)out.data_ptr(), N);
}

// Run a specific tuning variant (tools/gpu_conv_tune.py); returns false if
// the variant index does not match the geometry.
bool conv_fwd_variant(torch::Tensor in, torch::Tensor weight,
                      torch::Tensor bias, torch::Tensor out, int64_t stride,
                      int64_t variant) {
  const int N = (int)in.size(0), C = (int)in.size(1), H = (int)in.size(2),
            W = (int)in.size(3);
  const int COUT = (int)weight.size(0), KH = (int)weight.size(2),
            KW = (int)weight.size(3);
  const bool u8 = in.scalar_type() == torch::kUInt8;
  if (variant < 0 || variant >= (int64_t)(sizeof(kVariants) / sizeof(ConvLaunch)))
    return false;
  const ConvLaunch* L = &kVariants[variant];
  if (!(L->H == H && L->W == W && L->C == C && L->KH == KH && L->KW == KW &&
        L->S == (int)stride && L->COUT == COUT && L->u8 == u8))
    return false;
  const int P = (H - KH) / (int)stride + 1, Q = (W - KW) / (int)stride + 1;
  const int M = N * P * Q;
  const int rows_per_block = L->waves * L->rpw;
  const int blocks = (M + rows_per_block - 1) / rows_per_block;
  const __bf16* bias_ptr = nullptr;
  if (bias.defined() && bias.numel() > 0)
    bias_ptr = (const __bf16*)bias.data_ptr();
  hipLaunchKernelGGL(L->fn, dim3(blocks), dim3(L->waves * 64), L->lds_bytes,
                     (hipStream_t)at::cuda::getCurrentCUDAStream().stream(),
                     (const void*)in.data_ptr(),
                     (const __bf16*)weight.data_ptr(), bias_ptr,
                     (__bf16*)out.data_ptr(), N);
  return true;
}

void conv_wrw(torch::Tensor in, torch::Tensor gout, torch::Tensor gw_ws,
              torch::Tensor gb_ws, int64_t stride);  // defined below
bool conv_wrw_supported(int64_t H, int64_t W, int64_t C, int64_t KH,
                        int64_t KW, int64_t S, int64_t COUT, bool u8);
void conv_dgrad(torch::Tensor gout, torch::Tensor weight, torch::Tensor w_t,
                torch::Tensor dx, int64_t stride);  // defined below
bool conv_dgrad_supported(int64_t H, int64_t W, int64_t C, int64_t KH,
                          int64_t KW, int64_t S, int64_t COUT);
void conv_fwd_chw(torch::Tensor in, torch::Tensor weight, torch::Tensor bias,
                  torch::Tensor out2d, int64_t stride);  // defined below
bool conv_fwd_chw_supported(int64_t H, int64_t W, int64_t C, int64_t KH,
                            int64_t KW, int64_t S, int64_t COUT);
void relu_mask_bwd_chw(torch::Tensor gout2d, torch::Tensor out2d,
                       torch::Tensor dst_nhwc, int64_t C, int64_t PQ);
void linear_relu(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                 torch::Tensor out);  // defined below
bool linear_relu_supported(int64_t K, int64_t N);
void tr16_probe(torch::Tensor out, int64_t mode);

void register_conv(pybind11::module_& m) {
  m.def("conv_fwd", &conv_fwd,
        "fused NHWC bf16 MFMA conv fwd (+dequant on u8 input, bias, ReLU)");
  m.def("conv_fwd_supported", &conv_fwd_supported);
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA fragment-map probe");
  m.def("conv_fwd_variant", &conv_fwd_variant, "tuning-variant conv fwd");
  m.def("conv_wrw", &conv_wrw, "MFMA conv weight-grad (fp32 workspace)");
  m.def("conv_wrw_supported", &conv_wrw_supported);
  m.def("conv_dgrad", &conv_dgrad,
        "MFMA conv data-grad (masked-tap gather, pre-transposed weights)");
  m.def("conv_dgrad_supported", &conv_dgrad_supported);
  m.def("conv_fwd_chw", &conv_fwd_chw,
        "conv fwd with flatten fused into the epilogue (2-D CHW out)");
  m.def("conv_fwd_chw_supported", &conv_fwd_chw_supported);
  m.def("relu_mask_bwd_chw", &relu_mask_bwd_chw,
        "relu-mask + CHW->NHWC transpose (backward of the fused flatten)");
  m.def("linear_relu", &linear_relu,
        "fused Linear+bias+ReLU fwd (own MFMA GEMM, LDS-staged W)");
  m.def("linear_relu_supported", &linear_relu_supported);
  m.def("tr16_probe", &tr16_probe);
}

// ===========================================================================
// Weight-grad (wrw) kernel: gw[COUT][K] = sum_m gout^T[COUT][m] * im2col[m][K]
// — the tall-skinny reduction GEMM of conv backward (MIOpen's igemm_wrw was
// the single biggest post-fusion cost, profiles/). The reduction dim is the
// sample dim m, so both operands need m-major fragments: we stage 32-sample
// chunks into LDS TRANSPOSED (scalar b16 writes, padded rows) so fragment
// reads are clean b128s, MFMA accumulates COUT x 64-kelem tiles, and each
// block atomicAdds its partial into an fp32 workspace.
//   grid = (K/64 ktiles, MB m-slices); block = 4 waves; wave w owns the
//   16-kelem column block w.
// ===========================================================================

namespace {

constexpr int kPadM = 8;  // row pad (elements) for the staged tiles

typedef __bf16 v4bf_w __attribute__((ext_vector_type(4)));

// ds_read_tr16_b64 fragment read from a PLAIN row-major [32][NCOL+pad] tile:
// the 16-lane group transposes the 64 elements its lanes address, so with
// per-lane address base + (g*8 + (r>>2))*stride + 4*(r&3)  (g = lane>>4,
// r = lane&15) the result is exactly the MFMA operand fragment
// lane l -> elems j: tile[m = g*8 + j][col16 = r]  (j 0..3; second read at
// +4 rows gives j 4..7). Mapping verified on-GPU by tr16_probe (mode 1).
template <int STRIDE>
__device__ __forceinline__ bf16x8 tr_frag(const __bf16* tile, int lane,
                                          int colblk) {
  const int g = lane >> 4, r = lane & 15;
  const __bf16* base =
      tile + (g * 8 + (r >> 2)) * STRIDE + colblk * 16 + 4 * (r & 3);
  auto* p0 = (__attribute__((address_space(3))) v4bf_w*)base;
  auto* p1 = (__attribute__((address_space(3))) v4bf_w*)(base + 4 * STRIDE);
  v4bf_w lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p0);
  v4bf_w hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
  bf16x8 out;
#pragma unroll
  for (int j = 0; j < 4; ++j) { out[j] = lo[j]; out[j + 4] = hi[j]; }
  return out;
}

template <int H, int W, int C, int KH, int KW, int S, int COUT, bool U8IN>
__global__ __launch_bounds__(256) void conv_wrw_kernel(
    const void* __restrict__ in_v,      // (N,H,W,C) NHWC u8/bf16
    const __bf16* __restrict__ gout,    // (M, COUT) = NHWC grad (relu-masked)
    float* __restrict__ gw_ws,          // (COUT, K) fp32, pre-zeroed
    int batch, int mblocks) {
  using G = ConvGeom<H, W, C, KH, KW, S, COUT, U8IN>;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int ASTRIDE = 64 + kPadM;
  constexpr int GSTRIDE = COUT + kPadM;
  __bf16* a_t = reinterpret_cast<__bf16*>(smem);          // [32][ASTRIDE]
  __bf16* g_t = a_t + 32 * ASTRIDE;                       // [32][GSTRIDE]

  const int ktile = blockIdx.x;
  const int mb = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int M = batch * G::P * G::Q;
  const int n_chunks = (M + 31) / 32;

  constexpr int CBLK = COUT / 16;
  f32x4 acc[CBLK];
#pragma unroll
  for (int cb = 0; cb < CBLK; ++cb) acc[cb] = {0.f, 0.f, 0.f, 0.f};

  // staging assignment (round 2, ROUND2_PLAN item 3): the thread halves
  // stage a_t and g_t CONCURRENTLY — threads 0-127 load 16-elem pieces of
  // the im2col chunk (16 B/lane for u8 input, 2x16 B for bf16; was
  // 8 B/lane with all 256 threads serialized behind g_t), threads 128-255
  // stage gout. A 16-piece never crosses a patch row: ROWC % 16 == 0 for
  // every geometry.
  static_assert(G::ROWC % 16 == 0, "16-elem staging piece crosses a row");
  const int sm = tid >> 2;             // sample row 0..31 (tid < 128)
  const int sk16 = (tid & 3) * 16;     // kelem piece base 0..48

  for (int mc = mb; mc < n_chunks; mc += mblocks) {
    const int m0 = mc * 32;
    if (tid < 128) {
      // ---- stage im2col chunk -> a_t[m][kelem]
      const int m = m0 + sm;
      const int mcl = m < M ? m : M - 1;
      const int n = mcl / (G::P * G::Q);
      const int rem = mcl - n * (G::P * G::Q);
      const int p = rem / G::Q;
      const int q = rem - p * G::Q;
      const int kelem = ktile * 64 + sk16;
      const int dy = kelem / G::ROWC;
      const int dx = kelem - dy * G::ROWC;
      const int64_t goff =
          ((int64_t)n * H + p * S + dy) * (W * C) + q * S * C + dx;
      bf16x8 v0, v1;
      if constexpr (U8IN) {
        const uint8_t* src = reinterpret_cast<const uint8_t*>(in_v) + goff;
        uint4 raw = *reinterpret_cast<const uint4*>(src);
        const float inv255 = 1.0f / 255.0f;
        const unsigned rw[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          v0[j] = (__bf16)(((rw[j >> 2] >> (8 * (j & 3))) & 0xFF) * inv255);
          v1[j] = (__bf16)(((rw[2 + (j >> 2)] >> (8 * (j & 3))) & 0xFF)
                           * inv255);
        }
      } else {
        const __bf16* src = reinterpret_cast<const __bf16*>(in_v) + goff;
        v0 = *reinterpret_cast<const bf16x8*>(src);
        v1 = *reinterpret_cast<const bf16x8*>(src + 8);
      }
      if (m >= M) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          v0[j] = (__bf16)0.0f;
          v1[j] = (__bf16)0.0f;
        }
      }
      *reinterpret_cast<bf16x8*>(a_t + sm * ASTRIDE + sk16) = v0;
      *reinterpret_cast<bf16x8*>(a_t + sm * ASTRIDE + sk16 + 8) = v1;
    } else {
      // ---- stage gout chunk -> g_t[m][cout] (coalesced, zero-padded)
      constexpr int PIECES = COUT / 8;
      for (int piece = tid - 128; piece < 32 * PIECES; piece += 128) {
        const int m = m0 + piece / PIECES;
        const int c0 = (piece % PIECES) * 8;
        bf16x8 v;
        if (m < M) {
          v = *reinterpret_cast<const bf16x8*>(gout + (int64_t)m * COUT + c0);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) v[j] = (__bf16)0.0f;
        }
        *reinterpret_cast<bf16x8*>(g_t + (piece / PIECES) * GSTRIDE + c0) = v;
      }
    }
    __syncthreads();
    // ---- fragments via tr16 reads + MFMA: wave w owns kelem block w
    bf16x8 bfrag = tr_frag<ASTRIDE>(a_t, lane, wave);
#pragma unroll
    for (int cb = 0; cb < CBLK; ++cb) {
      bf16x8 afrag = tr_frag<GSTRIDE>(g_t, lane, cb);
      acc[cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[cb],
                                                        0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: atomic-accumulate partials into the fp32 workspace
  const int kout = ktile * 64 + wave * 16 + (lane & 15);
#pragma unroll
  for (int cb = 0; cb < CBLK; ++cb)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int cout = cb * 16 + (lane >> 4) * 4 + r;
      atomicAdd(&gw_ws[(int64_t)cout * G::K + kout], acc[cb][r]);
    }
}

struct WrwLaunch {
  int H, W, C, KH, KW, S, COUT;
  bool u8;
  void (*fn)(const void*, const __bf16*, float*, int, int);
  int lds_bytes;
  int ktiles;
};

template <int H, int W, int C, int KH, int KW, int S, int COUT, bool U8>
WrwLaunch make_wrw() {
  using G = ConvGeom<H, W, C, KH, KW, S, COUT, U8>;
  static_assert(G::K % 64 == 0, "wrw needs K % 64 == 0");
  return WrwLaunch{H, W, C, KH, KW, S, COUT, U8,
                   conv_wrw_kernel<H, W, C, KH, KW, S, COUT, U8>,
                   (int)(32 * (64 + kPadM + COUT + kPadM) * sizeof(__bf16)),
                   G::K / 64};
}

static const WrwLaunch kWrwLaunches[] = {
    make_wrw<84, 84, 4, 8, 8, 4, 32, true>(),
    make_wrw<84, 84, 4, 8, 8, 4, 32, false>(),
    make_wrw<20, 20, 32, 4, 4, 2, 64, false>(),
    make_wrw<9, 9, 64, 3, 3, 1, 64, false>(),
    make_wrw<84, 84, 4, 8, 8, 4, 16, true>(),
    make_wrw<84, 84, 4, 8, 8, 4, 16, false>(),
    make_wrw<20, 20, 16, 4, 4, 2, 32, false>(),
};

// bias grad: column sum of the (M, COUT) relu-masked gout — torch's
// strided bf16 reduce on channels_last costs ~11 us; this is one pass with
// a block-level LDS reduction and COUT atomics per block.
template <int COUT>
__global__ __launch_bounds__(256) void colsum_bf16_kernel(
    const __bf16* __restrict__ x, int64_t M, float* __restrict__ out) {
  // b128 loads (8 bf16/lane): COUT % 8 == 0, so a 16 B chunk never crosses a
  // row and covers columns phase..phase+7; the flat chunk stride
  // (gridDim*256*8) is a multiple of COUT, so each lane's phase is constant
  // and its 8 fp32 accumulators map to fixed columns. (Scalar 2 B loads
  // measured only ~0.5 TB/s — load-width bound.)
  typedef __bf16 bf8 __attribute__((ext_vector_type(8)));
  __shared__ float sh[COUT];
  const int tid = threadIdx.x;
  for (int c = tid; c < COUT; c += 256) sh[c] = 0.0f;
  __syncthreads();
  const int64_t nchunk = M * COUT / 8;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + tid;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int phase = (int)(((uint64_t)i * 8) % COUT);
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  const bf8* __restrict__ x8 = (const bf8*)x;
  for (; i < nchunk; i += stride) {
    bf8 v = x8[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += (float)v[j];
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) atomicAdd(&sh[phase + j], acc[j]);
  __syncthreads();
  if (tid < COUT) atomicAdd(&out[tid], sh[tid]);
}

}  // namespace

bool conv_wrw_supported(int64_t H, int64_t W, int64_t C, int64_t KH, int64_t KW,
                        int64_t S, int64_t COUT, bool u8) {
  for (const auto& l : kWrwLaunches)
    if (l.H == H && l.W == W && l.C == C && l.KH == KH && l.KW == KW &&
        l.S == S && l.COUT == COUT && l.u8 == u8)
      return true;
  return false;
}

// in: (N,C,H,W) channels_last u8/bf16; gout: (N,COUT,P,Q) channels_last bf16
// (already relu-masked); gw_ws: (COUT, KH*KW*C) fp32 pre-zeroed.
void conv_wrw(torch::Tensor in, torch::Tensor gout, torch::Tensor gw_ws,
              torch::Tensor gb_ws, int64_t stride) {
  TORCH_CHECK(in.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(gout.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(gw_ws.scalar_type() == torch::kFloat32 && gw_ws.is_contiguous());
  const int N = (int)in.size(0), C = (int)in.size(1), H = (int)in.size(2),
            W = (int)in.size(3);
  const int COUT = (int)gout.size(1);
  const bool u8 = in.scalar_type() == torch::kUInt8;
  const WrwLaunch* L = nullptr;
  for (const auto& l : kWrwLaunches) {
    int KHl = l.KH, KWl = l.KW;
    if (l.H == H && l.W == W && l.C == C && l.S == (int)stride &&
        l.COUT == COUT && l.u8 == u8 &&
        (H - KHl) / (int)stride + 1 == (int)gout.size(2) &&
        (W - KWl) / (int)stride + 1 == (int)gout.size(3)) {
      L = &l;
      break;
    }
  }
  TORCH_CHECK(L, "no wrw kernel for this geometry");
  const int P = (int)gout.size(2), Q = (int)gout.size(3);
  const int M = N * P * Q;
  const int n_chunks = (M + 31) / 32;
  // scale the m-grid with the batch: R2D2/IMPALA run 8-20x Ape-X's row
  // count through the same kernel (conv1 wrw measured 139 us at 1.02M
  // rows with 512 blocks = 250 serial chunks per block)
  int mb = (n_chunks > 8192 ? 2048 : 512) / L->ktiles;  // 4096 A/B'd worse (atomics)
  if (mb > n_chunks) mb = n_chunks;
  if (mb < 1) mb = 1;
  hipLaunchKernelGGL(L->fn, dim3(L->ktiles, mb), dim3(256), L->lds_bytes,
                     (hipStream_t)at::cuda::getCurrentCUDAStream().stream(),
                     (const void*)in.data_ptr(), (const __bf16*)gout.data_ptr(),
                     gw_ws.data_ptr<float>(), N, mb);
  if (gb_ws.defined() && gb_ws.numel() > 0) {
    const int64_t Mrows = (int64_t)N * P * Q;
    auto st = (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
    switch (COUT) {
      case 16:
        hipLaunchKernelGGL(colsum_bf16_kernel<16>, dim3(256), dim3(256), 0, st,
                           (const __bf16*)gout.data_ptr(), Mrows,
                           gb_ws.data_ptr<float>());
        break;
      case 32:
        hipLaunchKernelGGL(colsum_bf16_kernel<32>, dim3(256), dim3(256), 0, st,
                           (const __bf16*)gout.data_ptr(), Mrows,
                           gb_ws.data_ptr<float>());
        break;
      case 64:
        hipLaunchKernelGGL(colsum_bf16_kernel<64>, dim3(256), dim3(256), 0, st,
                           (const __bf16*)gout.data_ptr(), Mrows,
                           gb_ws.data_ptr<float>());
        break;
      default:
        TORCH_CHECK(false, "no colsum instantiation for COUT=", COUT);
    }
  }
}

// tr16 semantics probe: fill LDS with 0..511, each lane passes base +
// per-lane offset `mode`, dump what each lane's 4 elements are.
namespace {
typedef __bf16 v4bf __attribute__((ext_vector_type(4)));
__global__ void tr16_probe_kernel(float* out, int mode) {
  __shared__ __bf16 lds[512];
  int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 512; i += 64) lds[i] = (__bf16)(float)i;
  __syncthreads();
  int off;
  switch (mode) {
    case 0: off = 0; break;                 // uniform base
    case 1: off = lane * 4; break;          // natural v4 stride
    case 2: off = (lane >> 4) * 64; break;  // group stride 64
    default: off = (lane & 15) * 4; break;
  }
  auto* p = (__attribute__((address_space(3))) v4bf*)(lds + off);
  v4bf v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
#pragma unroll
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = (float)v[j];
}
}  // namespace

void tr16_probe(torch::Tensor out, int64_t mode) {
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)at::cuda::getCurrentCUDAStream().stream(),
                     out.data_ptr<float>(), (int)mode);
}

// ===========================================================================
// Data-grad (dgrad) kernel — retires MIOpen igemm_bwd / ck grouped-bwd from
// the hot path (VERDICT r01 missing-5). GEMM view:
//   dx[M=N*H*W rows][C cols] = sum_{kh,kw} G[(n,y,x)][COUT] x Wt[COUT][C]
// where G gathers gout[n,(y-kh)/S,(x-kw)/S,:] when the tap is valid (stride
// divisibility + bounds) and zero otherwise. K = KH*KW*COUT runs in 32-wide
// chunks that never straddle taps (COUT % 32 == 0 for every geometry).
// B-operand comes from a PRE-TRANSPOSED weight tensor
//   w_t[c][(kh*KW+kw)*COUT + cout] = w[cout][(kh*KW+kw)*C + c]
// (built once per step by conv_w_transpose below, L2-resident), so both
// operands are plain 16 B/lane loads — no LDS at all.
// ===========================================================================
namespace {

template <int H, int W, int C, int KH, int KW, int S, int COUT, int WAVES,
          int RPW>
__global__ __launch_bounds__(WAVES * 64) void conv_dgrad_kernel(
    const __bf16* __restrict__ gout,  // (N,P,Q,COUT) NHWC
    const __bf16* __restrict__ w_t,   // (C, KH*KW*COUT)
    __bf16* __restrict__ dx,          // (N,H,W,C) NHWC
    int batch) {
  constexpr int P = (H - KH) / S + 1;
  constexpr int Q = (W - KW) / S + 1;
  constexpr int K = KH * KW * COUT;
  constexpr int KCHUNKS = K / 32;
  constexpr int CPT = COUT / 32;  // chunks per tap
  static_assert(COUT % 32 == 0, "chunks must not straddle taps");
  constexpr int RFRAG = RPW / 16;
  constexpr int NFRAG = C / 16;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int M = batch * H * W;
  const int row0 = (blockIdx.x * WAVES + wave) * RPW;
  if (row0 >= M) return;

  // per-row-fragment (n, y, x) of this lane's A row
  int a_n[RFRAG], a_y[RFRAG], a_x[RFRAG];
#pragma unroll
  for (int rf = 0; rf < RFRAG; ++rf) {
    int arow = row0 + rf * 16 + (lane & 15);
    if (arow >= M) arow = M - 1;
    a_n[rf] = arow / (H * W);
    const int rem = arow - a_n[rf] * (H * W);
    a_y[rf] = rem / W;
    a_x[rf] = rem - a_y[rf] * W;
  }
  const int kpart = (lane >> 4) * 8;

  f32x4 acc[RFRAG][NFRAG];
#pragma unroll
  for (int rf = 0; rf < RFRAG; ++rf)
#pragma unroll
    for (int f = 0; f < NFRAG; ++f) acc[rf][f] = {0.f, 0.f, 0.f, 0.f};

#pragma unroll
  for (int kc = 0; kc < KCHUNKS; ++kc) {
    const int tap = kc / CPT;
    const int kh = tap / KW;
    const int kw = tap - kh * KW;
    const int co = (kc - tap * CPT) * 32 + kpart;
    bf16x8 a[RFRAG];
#pragma unroll
    for (int rf = 0; rf < RFRAG; ++rf) {
      const int py = a_y[rf] - kh;
      const int qx = a_x[rf] - kw;
      const int p = py / S, q = qx / S;
      const bool ok = py >= 0 && qx >= 0 && py == p * S && qx == q * S &&
                      p < P && q < Q;
      if (ok) {
        a[rf] = *reinterpret_cast<const bf16x8*>(
            gout + (((int64_t)a_n[rf] * P + p) * Q + q) * COUT + co);
      } else {
        a[rf] = bf16x8{};
      }
    }
#pragma unroll
    for (int f = 0; f < NFRAG; ++f) {
      const int col = f * 16 + (lane & 15);
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          w_t + (int64_t)col * K + tap * COUT + co);
#pragma unroll
      for (int rf = 0; rf < RFRAG; ++rf)
        acc[rf][f] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[rf], b, acc[rf][f], 0, 0, 0);
    }
  }

  const int crow_base = (lane >> 4) * 4;
#pragma unroll
  for (int rf = 0; rf < RFRAG; ++rf)
#pragma unroll
    for (int f = 0; f < NFRAG; ++f) {
      const int col = f * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int orow = row0 + rf * 16 + crow_base + r;
        if (orow < M) dx[(int64_t)orow * C + col] = (__bf16)acc[rf][f][r];
      }
    }
}

__global__ void conv_w_transpose_kernel(const __bf16* __restrict__ w,
                                        __bf16* __restrict__ w_t, int TAPS,
                                        int C, int COUT) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  const int total = TAPS * C * COUT;
  if (i >= total) return;
  const int co = i / (TAPS * C);
  const int rem = i - co * (TAPS * C);
  const int tap = rem / C;
  const int c = rem - tap * C;
  w_t[((int64_t)c * TAPS + tap) * COUT + co] = w[i];
}

struct DgradLaunch {
  int H, W, C, KH, KW, S, COUT;
  void (*fn)(const __bf16*, const __bf16*, __bf16*, int);
  int waves, rpw;
};

template <int H, int W, int C, int KH, int KW, int S, int COUT, int WAVES = 8,
          int RPW = 16>
DgradLaunch make_dgrad() {
  return DgradLaunch{H, W, C, KH, KW, S, COUT,
                     conv_dgrad_kernel<H, W, C, KH, KW, S, COUT, WAVES, RPW>,
                     WAVES, RPW};
}

static const DgradLaunch kDgrad[] = {
    make_dgrad<20, 20, 32, 4, 4, 2, 64>(),  // apex/r2d2 conv2
    make_dgrad<9, 9, 64, 3, 3, 1, 64>(),    // apex/r2d2 conv3
    make_dgrad<20, 20, 16, 4, 4, 2, 32>(),  // impala conv2
};
}  // namespace

bool conv_dgrad_supported(int64_t H, int64_t W, int64_t C, int64_t KH,
                          int64_t KW, int64_t S, int64_t COUT) {
  for (const auto& l : kDgrad)
    if (l.H == H && l.W == W && l.C == C && l.KH == KH && l.KW == KW &&
        l.S == S && l.COUT == COUT)
      return true;
  return false;
}

// gout: (N,COUT,P,Q) logical channels_last bf16; weight (COUT,C,KH,KW)
// logical channels_last bf16; dx out (N,C,H,W) logical channels_last bf16;
// w_t workspace (C*KH*KW*COUT) bf16.
void conv_dgrad(torch::Tensor gout, torch::Tensor weight, torch::Tensor w_t,
                torch::Tensor dx, int64_t stride) {
  TORCH_CHECK(gout.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(dx.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(weight.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int N = (int)dx.size(0), C = (int)dx.size(1), H = (int)dx.size(2),
            W = (int)dx.size(3);
  const int COUT = (int)weight.size(0), KH = (int)weight.size(2),
            KW = (int)weight.size(3);
  const DgradLaunch* L = nullptr;
  for (const auto& l : kDgrad)
    if (l.H == H && l.W == W && l.C == C && l.KH == KH && l.KW == KW &&
        l.S == (int)stride && l.COUT == COUT) {
      L = &l;
      break;
    }
  TORCH_CHECK(L, "no dgrad kernel for this geometry");
  auto stream = (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
  const int total = KH * KW * C * COUT;
  hipLaunchKernelGGL(conv_w_transpose_kernel, dim3((total + 255) / 256),
                     dim3(256), 0, stream, (const __bf16*)weight.data_ptr(),
                     (__bf16*)w_t.data_ptr(), KH * KW, C, COUT);
  const int M = N * H * W;
  const int rows_per_block = L->waves * L->rpw;
  const int blocks = (M + rows_per_block - 1) / rows_per_block;
  hipLaunchKernelGGL(L->fn, dim3(blocks), dim3(L->waves * 64), 0, stream,
                     (const __bf16*)gout.data_ptr(),
                     (const __bf16*)w_t.data_ptr(), (__bf16*)dx.data_ptr(), N);
}

// conv_fwd with the flatten fused into the epilogue (CHW-ordered 2-D out).
bool conv_fwd_chw_supported(int64_t H, int64_t W, int64_t C, int64_t KH,
                            int64_t KW, int64_t S, int64_t COUT) {
  for (const auto& l : kChwLaunches)
    if (l.H == H && l.W == W && l.C == C && l.KH == KH && l.KW == KW &&
        l.S == S && l.COUT == COUT)
      return true;
  return false;
}

void conv_fwd_chw(torch::Tensor in, torch::Tensor weight, torch::Tensor bias,
                  torch::Tensor out2d, int64_t stride) {
  TORCH_CHECK(in.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(weight.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(out2d.is_contiguous() && out2d.dim() == 2);
  const int N = (int)in.size(0), C = (int)in.size(1), H = (int)in.size(2),
            W = (int)in.size(3);
  const int COUT = (int)weight.size(0), KH = (int)weight.size(2),
            KW = (int)weight.size(3);
  const ConvLaunch* L = nullptr;
  for (const auto& l : kChwLaunches)
    if (l.H == H && l.W == W && l.C == C && l.KH == KH && l.KW == KW &&
        l.S == (int)stride && l.COUT == COUT) {
      L = &l;
      break;
    }
  TORCH_CHECK(L, "no CHW-out conv kernel for this geometry");
  const int P = (H - KH) / (int)stride + 1, Q = (W - KW) / (int)stride + 1;
  const int M = N * P * Q;
  const int rows_per_block = L->waves * L->rpw;
  const int blocks = (M + rows_per_block - 1) / rows_per_block;
  const __bf16* bias_ptr = nullptr;
  if (bias.defined() && bias.numel() > 0)
    bias_ptr = (const __bf16*)bias.data_ptr();
  hipLaunchKernelGGL(L->fn, dim3(blocks), dim3(L->waves * 64), L->lds_bytes,
                     (hipStream_t)at::cuda::getCurrentCUDAStream().stream(),
                     (const void*)in.data_ptr(),
                     (const __bf16*)weight.data_ptr(), bias_ptr,
                     (__bf16*)out2d.data_ptr(), N);
}

// relu-mask + CHW->NHWC transpose in one pass (backward of the fused
// flatten): masked[n,p,q,c] = gout_chw[n,c,pq] * (out_chw > 0)
namespace {
__global__ void relu_mask_bwd_chw_kernel(const __bf16* __restrict__ gout,
                                         const __bf16* __restrict__ outv,
                                         __bf16* __restrict__ dst, int64_t N,
                                         int C, int PQ) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t total = N * (int64_t)C * PQ;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int c = (int)(i % C);
    const int64_t t = i / C;
    const int pq = (int)(t % PQ);
    const int64_t n = t / PQ;
    const int64_t src = ((int64_t)n * C + c) * PQ + pq;
    dst[i] = ((float)outv[src] > 0.0f) ? gout[src] : (__bf16)0.0f;
  }
}
}  // namespace

void relu_mask_bwd_chw(torch::Tensor gout2d, torch::Tensor out2d,
                       torch::Tensor dst_nhwc, int64_t C, int64_t PQ) {
  const int64_t N = gout2d.size(0);
  const int64_t total = N * C * PQ;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(relu_mask_bwd_chw_kernel, dim3(blocks), dim3(256), 0,
                     (hipStream_t)at::cuda::getCurrentCUDAStream().stream(),
                     (const __bf16*)gout2d.data_ptr(),
                     (const __bf16*)out2d.data_ptr(),
                     (__bf16*)dst_nhwc.data_ptr(), N, (int)C, (int)PQ);
}

// ===========================================================================
// Fused Linear(+bias)+ReLU forward for the MLP trunk head (round 2):
// out(M, N) = relu(x(M, K) @ W(N, K)^T + b). hipBLASLt runs this shape
// (512, 3136)->(1024) at ~17.6 us + a separate 5 us ReLU; this kernel
// fuses the epilogue and tiles for the chip: 64 blocks x 4 waves, each
// wave a 16-row x 128-col strip, W k-slices staged through LDS per
// 64-wide K chunk (b128 fragment reads, conflict pad).
// ===========================================================================
namespace {

template <int K, int N, int NT /*cols per block*/>
__global__ __launch_bounds__(256) void linear_relu_kernel(
    const __bf16* __restrict__ x,   // (M, K)
    const __bf16* __restrict__ w,   // (N, K)
    const __bf16* __restrict__ b,   // (N) or null
    __bf16* __restrict__ out,       // (M, N)
    int M) {
  constexpr int KB = 64;            // K chunk staged per iteration
  constexpr int LROW = KB + 32;     // (LROW*2)%256 == 192: bank-spread pad
  __shared__ __bf16 wlds[NT][LROW];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  // block tile: 32 rows x NT cols; waves 2x2 over (16 rows x NT/2 cols)
  const int row0 = blockIdx.x * 32 + (wave >> 1) * 16;
  const int col0 = blockIdx.y * NT + (wave & 1) * (NT / 2);
  constexpr int NFRAG = NT / 2 / 16;
  const int arow = row0 + (lane & 15);
  const int arow_c = arow < M ? arow : (M > 0 ? M - 1 : 0);
  const int kpart = (lane >> 4) * 8;

  f32x4 acc[NFRAG];
#pragma unroll
  for (int f = 0; f < NFRAG; ++f) acc[f] = {0.f, 0.f, 0.f, 0.f};

  // software pipeline: the next chunk's W pieces are prefetched into
  // registers while the current chunk's MFMAs run (a single-buffered
  // stage->sync->MFMA loop exposed the full global latency per chunk)
  constexpr int PIECES = NT * KB / (256 * 8);
  bf16x8 wpre[PIECES];
#pragma unroll
  for (int pc = 0; pc < PIECES; ++pc) {
    const int base = (tid + pc * 256) * 8;
    const int r = base / KB;
    wpre[pc] = *reinterpret_cast<const bf16x8*>(
        w + (int64_t)(blockIdx.y * NT + r) * K + (base - r * KB));
  }
  // A fragments also 1-deep prefetched (they were the remaining serial
  // latency chain: one L2 round-trip per 16 MFMAs)
  bf16x8 a_cur[2], a_nxt[2];
#pragma unroll
  for (int kc = 0; kc < 2; ++kc)
    a_cur[kc] = *reinterpret_cast<const bf16x8*>(
        x + (int64_t)arow_c * K + kc * 32 + kpart);
  for (int k0 = 0; k0 < K; k0 += KB) {
    __syncthreads();
#pragma unroll
    for (int pc = 0; pc < PIECES; ++pc) {
      const int base = (tid + pc * 256) * 8;
      const int r = base / KB;
      *reinterpret_cast<bf16x8*>(&wlds[r][base - r * KB]) = wpre[pc];
    }
    __syncthreads();
    if (k0 + KB < K) {
#pragma unroll
      for (int kc = 0; kc < 2; ++kc)
        a_nxt[kc] = *reinterpret_cast<const bf16x8*>(
            x + (int64_t)arow_c * K + k0 + KB + kc * 32 + kpart);
    }
    if (k0 + KB < K) {
#pragma unroll
      for (int pc = 0; pc < PIECES; ++pc) {
        const int base = (tid + pc * 256) * 8;
        const int r = base / KB;
        wpre[pc] = *reinterpret_cast<const bf16x8*>(
            w + (int64_t)(blockIdx.y * NT + r) * K + k0 + KB +
            (base - r * KB));
      }
    }
#pragma unroll
    for (int kc = 0; kc < KB / 32; ++kc) {
      bf16x8 a = a_cur[kc];
#pragma unroll
      for (int f = 0; f < NFRAG; ++f) {
        const int wl = (wave & 1) * (NT / 2) + f * 16 + (lane & 15);
        bf16x8 bf = *reinterpret_cast<const bf16x8*>(
            &wlds[wl][kc * 32 + kpart]);
        acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bf, acc[f], 0, 0, 0);
      }
    }
    a_cur[0] = a_nxt[0];
    a_cur[1] = a_nxt[1];
  }
  const int crow = (lane >> 4) * 4;
#pragma unroll
  for (int f = 0; f < NFRAG; ++f) {
    const int col = col0 + f * 16 + (lane & 15);
    const float bv = b ? (float)b[col] : 0.0f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = row0 + crow + r;
      if (orow < M) {
        float v = acc[f][r] + bv;
        out[(int64_t)orow * N + col] = (__bf16)(v > 0.0f ? v : 0.0f);
      }
    }
  }
}
}  // namespace

bool linear_relu_supported(int64_t K, int64_t N) {
  return (K == 3136 && N == 1024);
}

void linear_relu(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                 torch::Tensor out) {
  const int M = (int)x.size(0), K = (int)x.size(1), N = (int)w.size(0);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(linear_relu_supported(K, N), "no linear_relu instantiation");
  const __bf16* bp =
      (b.defined() && b.numel()) ? (const __bf16*)b.data_ptr() : nullptr;
  dim3 grid((M + 31) / 32, N / 256);
  hipLaunchKernelGGL((linear_relu_kernel<3136, 1024, 256>), grid, dim3(256),
                     0, (hipStream_t)at::cuda::getCurrentCUDAStream().stream(),
                     (const __bf16*)x.data_ptr(), (const __bf16*)w.data_ptr(),
                     bp, (__bf16*)out.data_ptr(), M);
}
