// MI355X (gfx950 / CDNA4) kernels for the distributed_rl_amd learner hot path.
//
// Kernel inventory (reference op chains cited per kernel):
//   K1  dequant_u8_bf16      uint8 -> bf16 /255           (APE_X/Learner.py:61-67)
//   K4  dqn_loss fwd/bwd     n-step double-DQN TD loss + priority + IS weight
//                                                          (APE_X/Learner.py:83-114)
//   K6  value_rescale h/h^-1                               (R2D2/Learner.py:22-35)
//   K7  seq_priority 0.9*max+0.1*mean                      (R2D2/Learner.py:175-181)
//   K8  vtrace reversed scan                               (IMPALA/Learner.py:176-213)
//   K10 sum-tree PER: update/sample/min                    (contract SURVEY §2.8)
//   K11 fused grad-norm + clip                             (R2D2/Learner.py:200-211)
//
// Design notes (CDNA4): wave64; block sizes are multiples of 64; elementwise
// kernels are vectorized to >=8 B/lane and grid-stride-capped (guide §6 G11/G13);
// the sum-tree is lock-free — leaf swap via atomicExch, ancestor fix-up via
// atomicAdd deltas, so concurrent ingest/update batches compose correctly.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#define DRL_CHECK_CUDA(x) TORCH_CHECK(x.is_cuda(), #x " must be a device tensor")
#define DRL_CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

namespace {

constexpr int kBlock = 256;

__host__ __device__ inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

inline int grid_for(int64_t work, int per_thread = 1) {
  int64_t blocks = ceil_div(work, (int64_t)kBlock * per_thread);
  // cap at 256 CU x 8 blocks, grid-stride the rest (guide G11)
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// ---------------------------------------------------------------------------
// K1: uint8 -> bf16 (x/255). 16 u8 in -> 32 B out per lane.
// ---------------------------------------------------------------------------

__global__ void dequant_u8_bf16_kernel(const uchar4* __restrict__ in,
                                       ushort2* __restrict__ out,
                                       int64_t n4) {
  const float inv255 = 1.0f / 255.0f;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    uchar4 v = in[i];
    __hip_bfloat162 lo = __float22bfloat162_rn({v.x * inv255, v.y * inv255});
    __hip_bfloat162 hi = __float22bfloat162_rn({v.z * inv255, v.w * inv255});
    out[2 * i] = *reinterpret_cast<ushort2*>(&lo);
    out[2 * i + 1] = *reinterpret_cast<ushort2*>(&hi);
  }
}

__global__ void dequant_u8_f32_kernel(const uchar4* __restrict__ in,
                                      float4* __restrict__ out, int64_t n4) {
  const float inv255 = 1.0f / 255.0f;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
    uchar4 v = in[i];
    out[i] = {v.x * inv255, v.y * inv255, v.z * inv255, v.w * inv255};
  }
}

// NCHW uint8 -> NHWC bf16 (channels_last), C=4: one lane per (n,h,w) pixel
// reads 4 plane-strided bytes, writes one 8-byte bf16x4. Feeding MIOpen/our
// convs channels_last removes the batched_transpose layout kernels that
// dominate the NCHW path (see profiles/).
__global__ void dequant_u8_bf16_nhwc_c4_kernel(const uint8_t* __restrict__ in,
                                               ushort4* __restrict__ out,
                                               int64_t n_pix, int64_t hw) {
  const float inv255 = 1.0f / 255.0f;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_pix;
       i += stride) {
    int64_t n = i / hw;
    int64_t p = i - n * hw;          // h*W + w
    const uint8_t* src = in + n * 4 * hw + p;
    __hip_bfloat162 lo = __float22bfloat162_rn(
        {src[0] * inv255, src[hw] * inv255});
    __hip_bfloat162 hi = __float22bfloat162_rn(
        {src[2 * hw] * inv255, src[3 * hw] * inv255});
    ushort2 l = *reinterpret_cast<ushort2*>(&lo);
    ushort2 h = *reinterpret_cast<ushort2*>(&hi);
    out[i] = {l.x, l.y, h.x, h.y};
  }
}

// ---------------------------------------------------------------------------
// K10: lock-free sum-tree. Layout: float tree[2P], P = pow2 >= capacity,
// root tree[1], leaves tree[P + i]. tree[1] is the running total priority.
// ---------------------------------------------------------------------------

// Two-phase update kills root contention: phase 1 propagates atomic deltas
// only through the wide bottom levels (node >= TOP, where random leaves
// rarely collide); phase 2 (one block) deterministically rebuilds the top
// TOP-1 nodes level by level. 512 concurrent updates on a 2^17 tree went
// 136us -> ~6us with this split (profiles/).
constexpr int64_t kTreeTop = 2048;

__global__ void sumtree_update_kernel(float* __restrict__ tree,
                                      const int64_t* __restrict__ idx,
                                      const float* __restrict__ prio,
                                      int m, int64_t P, int64_t top) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= m) return;
  int64_t leaf = P + idx[i];
  float newp = prio[i];
  float old = atomicExch(&tree[leaf], newp);  // duplicate-safe: each swap sees
  float delta = newp - old;                   // a consistent predecessor value
  if (delta == 0.0f) return;
  for (int64_t node = leaf >> 1; node >= top; node >>= 1)
    atomicAdd(&tree[node], delta);
}

__global__ void sumtree_rebuild_top_kernel(float* __restrict__ tree,
                                           int64_t top) {
  for (int64_t s = top >> 1; s >= 1; s >>= 1) {
    for (int64_t i = s + threadIdx.x; i < 2 * s; i += blockDim.x)
      tree[i] = tree[2 * i] + tree[2 * i + 1];
    __syncthreads();
  }
}

__device__ inline float hash01(unsigned long long seed, unsigned int i) {
  // splitmix64 counter hash -> [0,1)
  unsigned long long z = seed + 0x9E3779B97F4A7C15ull * (1ull + i);
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  z = z ^ (z >> 31);
  return (float)(z >> 40) * (1.0f / 16777216.0f);
}

__global__ void sumtree_sample_kernel(const float* __restrict__ tree, int64_t P,
                                      int64_t n_valid, int k,
                                      const unsigned long long* __restrict__ seed,
                                      int64_t* __restrict__ out_idx,
                                      float* __restrict__ out_prob) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= k) return;
  float total = tree[1];
  // stratified inverse-CDF: stratum i draws u in [i/k, (i+1)/k) * total
  float u = (i + hash01(seed[0], i)) * (total / k);
  int64_t node = 1;
  while (node < P) {
    int64_t l = node << 1;
    float lv = tree[l];
    if (u < lv) {
      node = l;
    } else {
      u -= lv;
      node = l + 1;
    }
  }
  int64_t li = node - P;
  if (li >= n_valid) li = n_valid - 1;  // float round-off guard at the tail
  out_idx[i] = li;
  out_prob[i] = tree[P + li] / total;
}

__global__ void leaf_min_pos_kernel(const float* __restrict__ tree, int64_t P,
                                    int64_t n_valid,
                                    unsigned int* __restrict__ out_bits) {
  // min over positive leaf priorities; positive IEEE754 floats compare as uints
  __shared__ unsigned int smin[kBlock / 64];
  float m = INFINITY;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_valid;
       i += stride) {
    float v = tree[P + i];
    if (v > 0.0f && v < m) m = v;
  }
  // wave64 reduce
  for (int off = 32; off > 0; off >>= 1)
    m = fminf(m, __shfl_down(m, off, 64));
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  if (lane == 0) smin[wave] = __float_as_uint(m);
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned int b = smin[0];
    for (int w = 1; w < blockDim.x / 64; ++w) b = min(b, smin[w]);
    atomicMin(out_bits, b);
  }
}

__global__ void per_weights_kernel(const float* __restrict__ prob,
                                   const unsigned int* __restrict__ min_bits,
                                   const float* __restrict__ tree,
                                   int64_t n_valid, float beta, int k,
                                   float* __restrict__ out_w) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= k) return;
  float total = tree[1];
  float min_prob = __uint_as_float(min_bits[0]) / total;
  float max_w = __powf(1.0f / (n_valid * min_prob), beta);
  out_w[i] = __powf(1.0f / (n_valid * fmaxf(prob[i], 1e-12f)), beta) / max_w;
}

__global__ void bump_seed_kernel(unsigned long long* seed) {
  if (threadIdx.x == 0 && blockIdx.x == 0)
    seed[0] = seed[0] * 6364136223846793005ull + 1442695040888963407ull;
}

// ---------------------------------------------------------------------------
// K4: fused n-step double-DQN TD loss. One lane per batch row (A is tiny).
//   target = r + gamma^n * Q_tgt(s', argmax_a Q_on(s',a)) * (1-done)
//   td     = clamp(target - Q_on(s,a), -1, 1)
//   prio   = (|td| + 1e-7)^alpha ;  loss = 0.5*mean(w * td^2)
// ---------------------------------------------------------------------------

template <typename T>
__device__ __forceinline__ float drl_ld(const T* p, int64_t i) {
  return (float)p[i];
}

template <typename T>
__global__ void dqn_loss_fwd_kernel(
    const T* __restrict__ q_s, const T* __restrict__ q_sp_on,
    const T* __restrict__ q_sp_tg, const int64_t* __restrict__ act,
    const float* __restrict__ rew, const float* __restrict__ done,
    const float* __restrict__ w, int B, int A, float gamma_n, float alpha,
    float* __restrict__ loss_out /*pre-zeroed scalar*/,
    float* __restrict__ prio_out, float* __restrict__ grad_coef,
    float* __restrict__ qmax_out /*pre-zeroed scalar: mean of row maxes*/) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  float contrib = 0.0f, qm = 0.0f;
  if (i < B) {
    const T* row = q_sp_on + (int64_t)i * A;
    int a_star = 0;
    float best = drl_ld(row, 0);
    for (int a = 1; a < A; ++a) {
      float v = drl_ld(row, a);
      if (v > best) { best = v; a_star = a; }
    }
    float target = rew[i] + gamma_n * drl_ld(q_sp_tg, (int64_t)i * A + a_star)
                                * (1.0f - done[i]);
    float q = drl_ld(q_s, (int64_t)i * A + act[i]);
    // value telemetry: mean over rows of max_a Q(s,a)
    const T* srow = q_s + (int64_t)i * A;
    float smax = drl_ld(srow, 0);
    for (int a = 1; a < A; ++a) smax = fmaxf(smax, drl_ld(srow, a));
    qm = smax / B;
    float raw = target - q;
    float td = fminf(1.0f, fmaxf(-1.0f, raw));
    prio_out[i] = __powf(fabsf(td) + 1e-7f, alpha);
    float in_range = (raw > -1.0f && raw < 1.0f) ? 1.0f : 0.0f;
    float invB = 1.0f / B;
    grad_coef[i] = w[i] * td * in_range * invB;  // dL/dQ(s,a) = -grad_coef
    contrib = 0.5f * w[i] * td * td * invB;
  }
  // wave reduce then one atomic per wave (guide G12)
  for (int off = 32; off > 0; off >>= 1) {
    contrib += __shfl_down(contrib, off, 64);
    qm += __shfl_down(qm, off, 64);
  }
  if ((threadIdx.x & 63) == 0) {
    if (contrib != 0.0f) atomicAdd(loss_out, contrib);
    if (qm != 0.0f) atomicAdd(qmax_out, qm);
  }
}

template <typename T>
__global__ void dqn_loss_bwd_kernel(const float* __restrict__ grad_coef,
                                    const int64_t* __restrict__ act,
                                    const float* __restrict__ gout,
                                    int B, int A, T* __restrict__ grad_q) {
  int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= B * A) return;
  int i = j / A;
  int a = j - i * A;
  grad_q[j] = (T)((a == (int)act[i]) ? -grad_coef[i] * gout[0] : 0.0f);
}

// ---------------------------------------------------------------------------
// K6: value rescale h(x) = sign(x)(sqrt(|x|+1)-1) + eps*x and inverse.
// ---------------------------------------------------------------------------

__global__ void value_rescale_kernel(const float* __restrict__ x,
                                     float* __restrict__ y, int64_t n, float eps) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float v = x[i];
    float s = v >= 0.0f ? 1.0f : -1.0f;
    y[i] = s * (sqrtf(fabsf(v) + 1.0f) - 1.0f) + eps * v;
  }
}

__global__ void inv_value_rescale_kernel(const float* __restrict__ x,
                                         float* __restrict__ y, int64_t n,
                                         float eps) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float v = x[i];
    float s = v >= 0.0f ? 1.0f : -1.0f;
    float t = sqrtf(1.0f + 4.0f * eps * (fabsf(v) + 1.0f + eps)) - 1.0f;
    y[i] = s * (t * t / (4.0f * eps * eps) - 1.0f);
  }
}

// ---------------------------------------------------------------------------
// K7: sequence priority eta-mix over (T, B) |td|: (eta*max + (1-eta)*mean)^alpha
// ---------------------------------------------------------------------------

__global__ void seq_priority_kernel(const float* __restrict__ td_abs, int T,
                                    int B, float eta, float alpha,
                                    float* __restrict__ out) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float mx = 0.0f, sum = 0.0f;
  for (int t = 0; t < T; ++t) {
    float v = td_abs[(int64_t)t * B + b];
    mx = fmaxf(mx, v);
    sum += v;
  }
  out[b] = __powf(eta * mx + (1.0f - eta) * (sum / T), alpha);
}

// ---------------------------------------------------------------------------
// K8: V-trace reversed scan — sequential in T, parallel in B.
// Matches IMPALA/Learner.py:151-213 clipping order:
//   rho_c = min(rho_bar, exp(t_logp - b_logp)); c = lam*min(c_bar, rho)
//   delta_t = rho_c*(r_t + gamma*V_{t+1} - V_t);  acc = delta + gamma*c*acc
//   vs_t = V_t + acc;  pg_adv_t = rho_c*(r_t + gamma*vs_{t+1} - V_t)
// ---------------------------------------------------------------------------

__global__ void vtrace_kernel(const float* __restrict__ b_logp,
                              const float* __restrict__ t_logp,
                              const float* __restrict__ rew,
                              const float* __restrict__ values,
                              const float* __restrict__ boot,
                              const float* __restrict__ not_done, int T, int B,
                              float gamma, float rho_bar, float c_bar, float lam,
                              float* __restrict__ vs,
                              float* __restrict__ pg_adv,
                              float* __restrict__ rho_out) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float bv = boot[b] * not_done[b];
  float acc = 0.0f;
  for (int t = T - 1; t >= 0; --t) {
    int64_t o = (int64_t)t * B + b;
    float vtp1 = (t == T - 1) ? bv : values[o + B];
    float rho = expf(t_logp[o] - b_logp[o]);
    float rho_c = fminf(rho, rho_bar);
    float c = lam * fminf(rho, c_bar);
    float delta = rho_c * (rew[o] + gamma * vtp1 - values[o]);
    acc = delta + gamma * c * acc;
    vs[o] = values[o] + acc;
    rho_out[o] = rho_c;
  }
  for (int t = 0; t < T; ++t) {
    int64_t o = (int64_t)t * B + b;
    float vstp1 = (t == T - 1) ? bv : vs[o + B];
    pg_adv[o] = rho_out[o] * (rew[o] + gamma * vstp1 - values[o]);
  }
}

// (B,T)-layout variant: inputs/outputs row-major (B, T) so the learner needs
// no .t().contiguous() round-trips, and behavior probabilities mu enter raw
// (log taken in-kernel) — replaces 7 transpose/log launches per IMPALA step.
// Uncoalesced per-lane stride T is irrelevant here: the whole working set is
// a few KB and lives in L2.
__global__ void vtrace_bt_kernel(const float* __restrict__ mu,
                                 const float* __restrict__ t_logp,
                                 const float* __restrict__ rew,
                                 const float* __restrict__ values,
                                 const float* __restrict__ boot,
                                 const float* __restrict__ not_done, int T,
                                 int B, float gamma, float rho_bar,
                                 float c_bar, float lam,
                                 float* __restrict__ vs,
                                 float* __restrict__ pg_adv) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  const int64_t base = (int64_t)b * T;
  float bv = boot[b] * not_done[b];
  float acc = 0.0f;
  for (int t = T - 1; t >= 0; --t) {
    int64_t o = base + t;
    float vtp1 = (t == T - 1) ? bv : values[o + 1];
    float rho = expf(t_logp[o]) / mu[o];
    float rho_c = fminf(rho, rho_bar);
    float c = lam * fminf(rho, c_bar);
    float delta = rho_c * (rew[o] + gamma * vtp1 - values[o]);
    acc = delta + gamma * c * acc;
    vs[o] = values[o] + acc;
  }
  for (int t = 0; t < T; ++t) {
    int64_t o = base + t;
    float vstp1 = (t == T - 1) ? bv : vs[o + 1];
    float rho_c = fminf(expf(t_logp[o]) / mu[o], rho_bar);
    pg_adv[o] = rho_c * (rew[o] + gamma * vstp1 - values[o]);
  }
}

// ---------------------------------------------------------------------------
// K11: fused grad-norm + clip over a flat fp32 buffer (two launches, zero
// host sync: the scale is computed on-device from the sq-sum scalar).
// ---------------------------------------------------------------------------

__global__ void sq_sum_kernel(const float* __restrict__ x, int64_t n,
                              float* __restrict__ out /*pre-zeroed*/) {
  // float4 loads (16 B/lane) + one LDS block reduce -> ONE atomic per block
  // (was: scalar loads + 4 wave atomics/block; measured 19.7 us on a 6.8 MB
  // flat grad — ~350 GB/s, load-width bound).
  float acc = 0.0f;
  const int64_t n4 = n >> 2;
  const float4* __restrict__ x4 = (const float4*)x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t gid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (int64_t i = gid; i < n4; i += stride) {
    float4 v = x4[i];
    acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
  }
  for (int64_t i = (n4 << 2) + gid; i < n; i += stride) acc += x[i] * x[i];
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  __shared__ float wsum[kBlock / 64];
  if ((threadIdx.x & 63) == 0) wsum[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.0f;
    for (int w = 0; w < kBlock / 64; ++w) s += wsum[w];
    atomicAdd(out, s);
  }
}

__global__ void clip_scale_kernel(float* __restrict__ x, int64_t n,
                                  const float* __restrict__ sqsum,
                                  float max_norm) {
  float norm = sqrtf(sqsum[0]);
  float s = norm > max_norm ? max_norm / (norm + 1e-6f) : 1.0f;
  if (s == 1.0f) return;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    x[i] *= s;
}

}  // namespace

// ===========================================================================
// Torch bindings
// ===========================================================================

static inline hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

void dequant(torch::Tensor src_u8, torch::Tensor dst) {
  DRL_CHECK_CUDA(src_u8);
  DRL_CHECK_CONTIG(src_u8);
  DRL_CHECK_CONTIG(dst);
  TORCH_CHECK(src_u8.scalar_type() == torch::kUInt8);
  int64_t n = src_u8.numel();
  TORCH_CHECK(n % 4 == 0, "element count must be a multiple of 4");
  TORCH_CHECK(dst.numel() == n);
  int64_t n4 = n / 4;
  if (dst.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(dequant_u8_bf16_kernel, dim3(grid_for(n4)), dim3(kBlock), 0,
                       cur_stream(), (const uchar4*)src_u8.data_ptr(),
                       (ushort2*)dst.data_ptr(), n4);
  } else if (dst.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(dequant_u8_f32_kernel, dim3(grid_for(n4)), dim3(kBlock), 0,
                       cur_stream(), (const uchar4*)src_u8.data_ptr(),
                       (float4*)dst.data_ptr(), n4);
  } else {
    TORCH_CHECK(false, "dst must be bf16 or f32");
  }
}

void dequant_nhwc(torch::Tensor src_u8, torch::Tensor dst) {
  // src: contiguous NCHW uint8, C==4; dst: channels_last bf16 (same shape)
  DRL_CHECK_CUDA(src_u8);
  DRL_CHECK_CONTIG(src_u8);
  TORCH_CHECK(src_u8.scalar_type() == torch::kUInt8);
  TORCH_CHECK(dst.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(src_u8.dim() == 4 && src_u8.size(1) == 4, "expects (N,4,H,W)");
  TORCH_CHECK(dst.is_contiguous(at::MemoryFormat::ChannelsLast),
              "dst must be channels_last");
  int64_t N = src_u8.size(0), H = src_u8.size(2), W = src_u8.size(3);
  int64_t hw = H * W, n_pix = N * hw;
  hipLaunchKernelGGL(dequant_u8_bf16_nhwc_c4_kernel, dim3(grid_for(n_pix)),
                     dim3(kBlock), 0, cur_stream(),
                     (const uint8_t*)src_u8.data_ptr(), (ushort4*)dst.data_ptr(),
                     n_pix, hw);
}

void sumtree_update(torch::Tensor tree, torch::Tensor idx, torch::Tensor prio,
                    int64_t P) {
  DRL_CHECK_CUDA(tree);
  TORCH_CHECK(idx.scalar_type() == torch::kInt64);
  TORCH_CHECK(prio.scalar_type() == torch::kFloat32);
  int m = (int)idx.numel();
  if (m == 0) return;
  int64_t top = P < kTreeTop ? P : kTreeTop;
  hipLaunchKernelGGL(sumtree_update_kernel, dim3(ceil_div(m, kBlock)), dim3(kBlock),
                     0, cur_stream(), tree.data_ptr<float>(),
                     idx.data_ptr<int64_t>(), prio.data_ptr<float>(), m, P, top);
  hipLaunchKernelGGL(sumtree_rebuild_top_kernel, dim3(1), dim3(1024), 0,
                     cur_stream(), tree.data_ptr<float>(), top);
}

void sumtree_sample(torch::Tensor tree, int64_t P, int64_t n_valid, int64_t k,
                    torch::Tensor seed, torch::Tensor out_idx,
                    torch::Tensor out_prob) {
  DRL_CHECK_CUDA(tree);
  hipLaunchKernelGGL(sumtree_sample_kernel, dim3(ceil_div(k, kBlock)), dim3(kBlock),
                     0, cur_stream(), tree.data_ptr<float>(), P, n_valid, (int)k,
                     (const unsigned long long*)seed.data_ptr(),
                     out_idx.data_ptr<int64_t>(), out_prob.data_ptr<float>());
}

void leaf_min_pos(torch::Tensor tree, int64_t P, int64_t n_valid,
                  torch::Tensor out_bits) {
  hipLaunchKernelGGL(leaf_min_pos_kernel, dim3(grid_for(n_valid, 8)), dim3(kBlock),
                     0, cur_stream(), tree.data_ptr<float>(), P, n_valid,
                     (unsigned int*)out_bits.data_ptr());
}

void per_weights(torch::Tensor prob, torch::Tensor min_bits, torch::Tensor tree,
                 int64_t n_valid, double beta, torch::Tensor out_w) {
  int k = (int)prob.numel();
  hipLaunchKernelGGL(per_weights_kernel, dim3(ceil_div(k, kBlock)), dim3(kBlock), 0,
                     cur_stream(), prob.data_ptr<float>(),
                     (const unsigned int*)min_bits.data_ptr(),
                     tree.data_ptr<float>(), n_valid, (float)beta, k,
                     out_w.data_ptr<float>());
}

void bump_seed(torch::Tensor seed) {
  hipLaunchKernelGGL(bump_seed_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     (unsigned long long*)seed.data_ptr());
}

void dqn_loss_fwd(torch::Tensor q_s, torch::Tensor q_sp_on, torch::Tensor q_sp_tg,
                  torch::Tensor act, torch::Tensor rew, torch::Tensor done,
                  torch::Tensor w, double gamma_n, double alpha,
                  torch::Tensor loss_out, torch::Tensor prio_out,
                  torch::Tensor grad_coef, torch::Tensor qmax_out) {
  int B = (int)q_s.size(0), A = (int)q_s.size(1);
  if (q_s.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(dqn_loss_fwd_kernel<__bf16>, dim3(ceil_div(B, kBlock)),
                       dim3(kBlock), 0, cur_stream(),
                       (const __bf16*)q_s.data_ptr(),
                       (const __bf16*)q_sp_on.data_ptr(),
                       (const __bf16*)q_sp_tg.data_ptr(),
                       act.data_ptr<int64_t>(), rew.data_ptr<float>(),
                       done.data_ptr<float>(), w.data_ptr<float>(), B, A,
                       (float)gamma_n, (float)alpha, loss_out.data_ptr<float>(),
                       prio_out.data_ptr<float>(), grad_coef.data_ptr<float>(),
                       qmax_out.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(dqn_loss_fwd_kernel<float>, dim3(ceil_div(B, kBlock)),
                       dim3(kBlock), 0, cur_stream(), q_s.data_ptr<float>(),
                       q_sp_on.data_ptr<float>(), q_sp_tg.data_ptr<float>(),
                       act.data_ptr<int64_t>(), rew.data_ptr<float>(),
                       done.data_ptr<float>(), w.data_ptr<float>(), B, A,
                       (float)gamma_n, (float)alpha, loss_out.data_ptr<float>(),
                       prio_out.data_ptr<float>(), grad_coef.data_ptr<float>(),
                       qmax_out.data_ptr<float>());
  }
}

void dqn_loss_bwd(torch::Tensor grad_coef, torch::Tensor act, torch::Tensor gout,
                  torch::Tensor grad_q) {
  int B = (int)grad_q.size(0), A = (int)grad_q.size(1);
  if (grad_q.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(dqn_loss_bwd_kernel<__bf16>,
                       dim3(ceil_div((int64_t)B * A, kBlock)), dim3(kBlock), 0,
                       cur_stream(), grad_coef.data_ptr<float>(),
                       act.data_ptr<int64_t>(), gout.data_ptr<float>(), B, A,
                       (__bf16*)grad_q.data_ptr());
  } else {
    hipLaunchKernelGGL(dqn_loss_bwd_kernel<float>,
                       dim3(ceil_div((int64_t)B * A, kBlock)), dim3(kBlock), 0,
                       cur_stream(), grad_coef.data_ptr<float>(),
                       act.data_ptr<int64_t>(), gout.data_ptr<float>(), B, A,
                       grad_q.data_ptr<float>());
  }
}

void value_rescale(torch::Tensor x, torch::Tensor y, double eps) {
  int64_t n = x.numel();
  hipLaunchKernelGGL(value_rescale_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                     cur_stream(), x.data_ptr<float>(), y.data_ptr<float>(), n,
                     (float)eps);
}

void inv_value_rescale(torch::Tensor x, torch::Tensor y, double eps) {
  int64_t n = x.numel();
  hipLaunchKernelGGL(inv_value_rescale_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                     cur_stream(), x.data_ptr<float>(), y.data_ptr<float>(), n,
                     (float)eps);
}

void seq_priority(torch::Tensor td_abs, double eta, double alpha,
                  torch::Tensor out) {
  int T = (int)td_abs.size(0), B = (int)td_abs.size(1);
  hipLaunchKernelGGL(seq_priority_kernel, dim3(ceil_div(B, kBlock)), dim3(kBlock), 0,
                     cur_stream(), td_abs.data_ptr<float>(), T, B, (float)eta,
                     (float)alpha, out.data_ptr<float>());
}

void vtrace(torch::Tensor b_logp, torch::Tensor t_logp, torch::Tensor rew,
            torch::Tensor values, torch::Tensor boot, torch::Tensor not_done,
            double gamma, double rho_bar, double c_bar, double lam,
            torch::Tensor vs, torch::Tensor pg_adv, torch::Tensor rho_out) {
  int T = (int)rew.size(0), B = (int)rew.size(1);
  hipLaunchKernelGGL(vtrace_kernel, dim3(ceil_div(B, kBlock)), dim3(kBlock), 0,
                     cur_stream(), b_logp.data_ptr<float>(),
                     t_logp.data_ptr<float>(), rew.data_ptr<float>(),
                     values.data_ptr<float>(), boot.data_ptr<float>(),
                     not_done.data_ptr<float>(), T, B, (float)gamma,
                     (float)rho_bar, (float)c_bar, (float)lam,
                     vs.data_ptr<float>(), pg_adv.data_ptr<float>(),
                     rho_out.data_ptr<float>());
}

void vtrace_bt(torch::Tensor mu, torch::Tensor t_logp, torch::Tensor rew,
               torch::Tensor values, torch::Tensor boot, torch::Tensor not_done,
               double gamma, double rho_bar, double c_bar, double lam,
               torch::Tensor vs, torch::Tensor pg_adv) {
  int B = (int)rew.size(0), T = (int)rew.size(1);
  for (auto* t : {&mu, &t_logp, &values, &vs, &pg_adv}) {
    DRL_CHECK_CONTIG((*t));
    TORCH_CHECK(t->numel() == (int64_t)B * T, "vtrace_bt shape mismatch");
  }
  TORCH_CHECK(boot.numel() == B && not_done.numel() == B);
  hipLaunchKernelGGL(vtrace_bt_kernel, dim3(ceil_div(B, kBlock)), dim3(kBlock),
                     0, cur_stream(), mu.data_ptr<float>(),
                     t_logp.data_ptr<float>(), rew.data_ptr<float>(),
                     values.data_ptr<float>(), boot.data_ptr<float>(),
                     not_done.data_ptr<float>(), T, B, (float)gamma,
                     (float)rho_bar, (float)c_bar, (float)lam,
                     vs.data_ptr<float>(), pg_adv.data_ptr<float>());
}

void grad_clip(torch::Tensor flat, double max_norm, torch::Tensor sqsum_buf) {
  int64_t n = flat.numel();
  hipLaunchKernelGGL(sq_sum_kernel, dim3(grid_for(n, 8)), dim3(kBlock), 0,
                     cur_stream(), flat.data_ptr<float>(), n,
                     sqsum_buf.data_ptr<float>());
  hipLaunchKernelGGL(clip_scale_kernel, dim3(grid_for(n, 4)), dim3(kBlock), 0,
                     cur_stream(), flat.data_ptr<float>(), n,
                     sqsum_buf.data_ptr<float>(), (float)max_norm);
}

void register_conv(pybind11::module_& m);  // conv_mfma.hip
void relu_mask_bwd(torch::Tensor gout, torch::Tensor out, torch::Tensor dst);
void dueling_fwd(torch::Tensor adv, torch::Tensor val, torch::Tensor out);
void dueling_bwd(torch::Tensor g, torch::Tensor gadv, torch::Tensor gval);
void lstm_cell_fwd(torch::Tensor gates, torch::Tensor c_prev,
                   torch::Tensor h_out, torch::Tensor c_out,
                   torch::Tensor acts, torch::Tensor tanhc);
void lstm_cell_bwd(torch::Tensor dh, torch::Tensor gout_t, torch::Tensor dc_in,
                   torch::Tensor acts, torch::Tensor tanhc,
                   torch::Tensor c_prev, torch::Tensor dgates,
                   torch::Tensor dc_prev);
void policy_loss_fwd(torch::Tensor logits, torch::Tensor act, torch::Tensor adv,
                     double er, torch::Tensor obj_out, torch::Tensor ent_out,
                     torch::Tensor logpa_out, torch::Tensor pi_save,
                     torch::Tensor H_save);
void policy_loss_bwd(torch::Tensor pi_save, torch::Tensor H_save,
                     torch::Tensor act, torch::Tensor adv, torch::Tensor gout,
                     double er, double gsign, torch::Tensor dlogits);
void impala_loss_fwd(torch::Tensor logpa, torch::Tensor adv,
                     torch::Tensor mean_H, torch::Tensor v, torch::Tensor vs,
                     double er, torch::Tensor loss_out, torch::Tensor obj_out,
                     torch::Tensor critic_out);
void impala_out_bwd(torch::Tensor pi_save, torch::Tensor H_save,
                    torch::Tensor act, torch::Tensor adv, torch::Tensor v,
                    torch::Tensor vs, torch::Tensor gloss, int64_t B,
                    int64_t T, int64_t A, double er, torch::Tensor dout);
void rmsprop_step(torch::Tensor p, torch::Tensor g, torch::Tensor sq,
                  torch::Tensor ga, torch::Tensor mom, double lr, double alpha,
                  double eps, double wd, double mu, bool centered, bool has_mom);
void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, torch::Tensor step, double lr, double b1,
               double b2, double eps, double wd);
bool lstm_step_fused(torch::Tensor xp_t, torch::Tensor h_in, torch::Tensor c_in,
                     torch::Tensor w_hh, torch::Tensor h_out,
                     torch::Tensor c_out, torch::Tensor acts,
                     torch::Tensor tanhc);
bool lstm_seq_persistent(torch::Tensor xp, torch::Tensor hs, torch::Tensor cs,
                         torch::Tensor w_hh, torch::Tensor acts,
                         torch::Tensor tanhc, torch::Tensor ctr);
bool lstm_step_fwd_bf16(torch::Tensor xp_t, torch::Tensor h_bf,
                        torch::Tensor c_in, torch::Tensor w_bf,
                        torch::Tensor h_out, torch::Tensor c_out,
                        torch::Tensor h_bf_out, torch::Tensor acts_t,
                        torch::Tensor tanhc_t);
bool lstm_step_bwd_bf16(torch::Tensor dg_prev, torch::Tensor dh_init,
                        torch::Tensor gout_t, torch::Tensor dc_in,
                        torch::Tensor w_t_bf, torch::Tensor acts_t,
                        torch::Tensor tanhc_t, torch::Tensor c_prev,
                        torch::Tensor dgates_t, torch::Tensor dg_bf_t,
                        torch::Tensor dc_out);
void dueling_dqn_loss_fwd(torch::Tensor adv_s, torch::Tensor val_s,
                          torch::Tensor adv_on, torch::Tensor val_on,
                          torch::Tensor adv_tg, torch::Tensor val_tg,
                          torch::Tensor act, torch::Tensor rew,
                          torch::Tensor done, torch::Tensor w, double gamma_n,
                          double alpha, torch::Tensor loss_out,
                          torch::Tensor prio_out, torch::Tensor grad_coef,
                          torch::Tensor qmax_out);
void dueling_dqn_loss_bwd(torch::Tensor grad_coef, torch::Tensor act,
                          torch::Tensor gout, torch::Tensor g_adv,
                          torch::Tensor g_val);
void pack_rows(torch::Tensor src, int64_t rec_size, std::vector<int64_t> offs,
               std::vector<int64_t> sizes, std::vector<torch::Tensor> dsts,
               int64_t dst_row);
bool lstm_seq_fwd_bf16(torch::Tensor xp, torch::Tensor c0, torch::Tensor w_bf,
                       torch::Tensor hs, torch::Tensor cs, torch::Tensor h_bfs,
                       torch::Tensor acts, torch::Tensor tanhc,
                       torch::Tensor ctr);
bool lstm_seq_bwd_bf16(torch::Tensor dh_init, torch::Tensor gout,
                       torch::Tensor dc_T, torch::Tensor w_t_bf,
                       torch::Tensor acts, torch::Tensor tanhc,
                       torch::Tensor cs, torch::Tensor dgates,
                       torch::Tensor dgates_bf, torch::Tensor dc0_out,
                       torch::Tensor ctr);
void dueling_q_loss_fwd(torch::Tensor h_s, torch::Tensor h_on,
                        torch::Tensor h_tg, torch::Tensor wa, torch::Tensor ba,
                        torch::Tensor wv, torch::Tensor bv, torch::Tensor wa_t,
                        torch::Tensor ba_t, torch::Tensor wv_t,
                        torch::Tensor bv_t, torch::Tensor act,
                        torch::Tensor rew, torch::Tensor done, torch::Tensor w,
                        double gamma_n, double alpha, torch::Tensor loss_out,
                        torch::Tensor prio_out, torch::Tensor grad_coef,
                        torch::Tensor qmax_out);
void dueling_q_loss_bwd(torch::Tensor grad_coef, torch::Tensor act,
                        torch::Tensor gout, torch::Tensor wa, torch::Tensor wv,
                        torch::Tensor h_s, torch::Tensor dh, torch::Tensor dwa,
                        torch::Tensor dba, torch::Tensor dwv,
                        torch::Tensor dbv);
void r2d2_loss_fwd(torch::Tensor q_train, torch::Tensor q_tgt,
                   torch::Tensor act, torch::Tensor rew, torch::Tensor done,
                   torch::Tensor w, int64_t m, int64_t n_step, double gamma,
                   bool rescale, torch::Tensor td_out, torch::Tensor stats);
void r2d2_prio(torch::Tensor td, double alpha, double eta, torch::Tensor prio);
void r2d2_loss_bwd(torch::Tensor td, torch::Tensor act, torch::Tensor w,
                   torch::Tensor gout, int64_t T, int64_t m, torch::Tensor dq);
void seq_transpose_rows(torch::Tensor src, torch::Tensor dst);
void gather_rows(torch::Tensor idx, std::vector<torch::Tensor> srcs,
                 std::vector<torch::Tensor> dsts);
void lstm_diag(torch::Tensor h_bfs, torch::Tensor w_bf, torch::Tensor sink,
               torch::Tensor ctr, int64_t B, int64_t T, int64_t mode);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  register_conv(m);
  m.def("dequant", &dequant, "u8 -> bf16/f32 /255 (K1)");
  m.def("dequant_nhwc", &dequant_nhwc, "u8 NCHW -> bf16 NHWC /255 (K1)");
  m.def("sumtree_update", &sumtree_update, "lock-free sum-tree leaf update (K10)");
  m.def("sumtree_sample", &sumtree_sample, "stratified sum-tree descent (K10)");
  m.def("leaf_min_pos", &leaf_min_pos, "min positive leaf priority (K10)");
  m.def("per_weights", &per_weights, "PER IS weights (K10)");
  m.def("bump_seed", &bump_seed, "advance device RNG seed (graph-safe)");
  m.def("dqn_loss_fwd", &dqn_loss_fwd, "fused n-step double-DQN loss fwd (K4)");
  m.def("dqn_loss_bwd", &dqn_loss_bwd, "fused n-step double-DQN loss bwd (K4)");
  m.def("value_rescale", &value_rescale, "R2D2 h(x) (K6)");
  m.def("inv_value_rescale", &inv_value_rescale, "R2D2 h^-1(x) (K6)");
  m.def("seq_priority", &seq_priority, "R2D2 eta-mix sequence priority (K7)");
  m.def("vtrace", &vtrace, "IMPALA V-trace reversed scan (K8)");
  m.def("vtrace_bt", &vtrace_bt,
        "V-trace on (B,T) row-major tensors, mu probs raw (K8, no transposes)");
  m.def("grad_clip", &grad_clip, "fused global grad-norm clip (K11)");
  m.def("relu_mask_bwd", &relu_mask_bwd, "dst = gout * (out > 0), bf16");
  m.def("dueling_fwd", &dueling_fwd, "fused (A+V)-mean(A) (K3)");
  m.def("dueling_bwd", &dueling_bwd, "dueling epilogue backward (K3)");
  m.def("lstm_cell_fwd", &lstm_cell_fwd, "fused LSTM cell forward (K5)");
  m.def("lstm_cell_bwd", &lstm_cell_bwd, "fused LSTM cell backward (K5)");
  m.def("policy_loss_fwd", &policy_loss_fwd, "fused IMPALA policy obj fwd (K9)");
  m.def("policy_loss_bwd", &policy_loss_bwd, "fused IMPALA policy obj bwd (K9)");
  m.def("impala_loss_fwd", &impala_loss_fwd,
        "fused IMPALA total loss: pg objective + entropy + critic MSE (K9)");
  m.def("impala_out_bwd", &impala_out_bwd,
        "IMPALA whole-head backward: d loss/d out in one launch (K9)");
  m.def("rmsprop_step", &rmsprop_step, "fused flat centered RMSprop (K12)");
  m.def("adam_step", &adam_step, "fused flat Adam (K12)");
  m.def("lstm_step_fused", &lstm_step_fused,
        "one-kernel LSTM timestep: fp32-MFMA hh GEMM + cell (K5)");
  m.def("lstm_seq_persistent", &lstm_seq_persistent,
        "whole-sequence persistent LSTM: grid-resident, agent-scope step "
        "barriers (K5)");
  m.def("lstm_step_fwd_bf16", &lstm_step_fwd_bf16,
        "one-kernel LSTM timestep fwd: bf16-MFMA hh GEMM + cell (K5 v2)");
  m.def("lstm_step_bwd_bf16", &lstm_step_bwd_bf16,
        "one-kernel LSTM timestep bwd: bf16-MFMA dh GEMM + cell-bwd (K5 v2)");
  m.def("dueling_dqn_loss_fwd", &dueling_dqn_loss_fwd,
        "whole-head dueling + n-step double-DQN loss fwd (K3+K4 fused)");
  m.def("dueling_dqn_loss_bwd", &dueling_dqn_loss_bwd,
        "whole-head dueling loss bwd: closed-form (g_adv, g_val) (K3+K4)");
  m.def("pack_rows", &pack_rows,
        "AoS records -> SoA pinned staging, GIL-free multithreaded (C2)",
        pybind11::call_guard<pybind11::gil_scoped_release>());
  m.def("lstm_seq_fwd_bf16", &lstm_seq_fwd_bf16,
        "persistent whole-sequence LSTM fwd, bf16-MFMA hh + grid barriers "
        "(K5 v3)");
  m.def("lstm_seq_bwd_bf16", &lstm_seq_bwd_bf16,
        "persistent whole-sequence LSTM bwd, bf16-MFMA dh + grid barriers "
        "(K5 v3)");
  m.def("dueling_q_loss_fwd", &dueling_q_loss_fwd,
        "heads+dueling+n-step-DQN loss in one kernel (K3+K4+heads)");
  m.def("dueling_q_loss_bwd", &dueling_q_loss_bwd,
        "closed-form dh + head weight/bias grads (2 kernels)");
  m.def("r2d2_loss_fwd", &r2d2_loss_fwd,
        "R2D2 n-step targets + rescale + IS loss, one kernel (K4/K6 seq)");
  m.def("r2d2_prio", &r2d2_prio, "eta-mix sequence priority (K7)");
  m.def("r2d2_loss_bwd", &r2d2_loss_bwd,
        "R2D2 loss backward: closed-form dq_train scatter");
  m.def("seq_transpose_rows", &seq_transpose_rows,
        "(B,T,row) -> (T,B,row) whole-row block copy");
  m.def("gather_rows", &gather_rows,
        "fused multi-column replay row gather (one launch per sample)");
  m.def("lstm_diag", &lstm_diag,
        "persistent-step cost decomposition (barrier/stage/mfma)");
}
// appended: fused ReLU-mask backward (gout *= (out > 0)), bf16, one pass —
// replaces the bool-compare + mul pair per conv layer in the fused-conv
// backward (profiles/: the elementwise glue was ~150 us/step).
namespace {
__global__ void relu_mask_bwd_kernel(const ushort4* __restrict__ out,
                                     const ushort4* __restrict__ gout,
                                     ushort4* __restrict__ dst, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    ushort4 o = out[i];
    ushort4 g = gout[i];
    // bf16 > 0 <=> sign bit clear and not zero
    g.x = (o.x & 0x8000 || o.x == 0) ? 0 : g.x;
    g.y = (o.y & 0x8000 || o.y == 0) ? 0 : g.y;
    g.z = (o.z & 0x8000 || o.z == 0) ? 0 : g.z;
    g.w = (o.w & 0x8000 || o.w == 0) ? 0 : g.w;
    dst[i] = g;
  }
}
}  // namespace

void relu_mask_bwd(torch::Tensor gout, torch::Tensor out, torch::Tensor dst) {
  TORCH_CHECK(gout.scalar_type() == torch::kBFloat16 &&
              out.scalar_type() == torch::kBFloat16);
  int64_t n = gout.numel();
  TORCH_CHECK(n % 4 == 0);
  hipLaunchKernelGGL(relu_mask_bwd_kernel, dim3(grid_for(n / 4)), dim3(kBlock),
                     0, cur_stream(), (const ushort4*)out.data_ptr(),
                     (const ushort4*)gout.data_ptr(), (ushort4*)dst.data_ptr(),
                     n / 4);
}

// K3: fused dueling-head epilogue  out = (A + V) - mean(A)
// (reference graph nodes Add/Mean/Substract, cfg/ape_x.json:72-88 — three
// eager kernels per forward collapse into one; backward likewise).
namespace {
__global__ void dueling_fwd_kernel(const float* __restrict__ adv,
                                   const float* __restrict__ val, int B, int A,
                                   float* __restrict__ out) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  const float* a = adv + (int64_t)i * A;
  float mean = 0.0f;
  for (int j = 0; j < A; ++j) mean += a[j];
  mean /= A;
  float v = val[i];
  float* o = out + (int64_t)i * A;
  for (int j = 0; j < A; ++j) o[j] = a[j] + v - mean;
}

__global__ void dueling_bwd_kernel(const float* __restrict__ g, int B, int A,
                                   float* __restrict__ gadv,
                                   float* __restrict__ gval) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  const float* gi = g + (int64_t)i * A;
  float s = 0.0f;
  for (int j = 0; j < A; ++j) s += gi[j];
  gval[i] = s;
  float m = s / A;
  float* ga = gadv + (int64_t)i * A;
  for (int j = 0; j < A; ++j) ga[j] = gi[j] - m;
}
}  // namespace

void dueling_fwd(torch::Tensor adv, torch::Tensor val, torch::Tensor out) {
  int B = (int)adv.size(0), A = (int)adv.size(1);
  hipLaunchKernelGGL(dueling_fwd_kernel, dim3(ceil_div(B, kBlock)), dim3(kBlock),
                     0, cur_stream(), adv.data_ptr<float>(),
                     val.data_ptr<float>(), B, A, out.data_ptr<float>());
}

void dueling_bwd(torch::Tensor g, torch::Tensor gadv, torch::Tensor gval) {
  int B = (int)g.size(0), A = (int)g.size(1);
  hipLaunchKernelGGL(dueling_bwd_kernel, dim3(ceil_div(B, kBlock)), dim3(kBlock),
                     0, cur_stream(), g.data_ptr<float>(), B, A,
                     gadv.data_ptr<float>(), gval.data_ptr<float>());
}

// K5-lite: fused LSTM cell pointwise (forward + backward). The sequence
// loop lives on the host (one hh-GEMM + one cell kernel per step, all
// fixed-shape -> hipGraph-capturable, unlike MIOpen's RNN path); the
// input projection is hoisted into one big GEMM over all timesteps.
// Gate order matches nn.LSTM: [i | f | g | o].
namespace {
__global__ void lstm_cell_fwd_kernel(
    const float* __restrict__ gates,   // (B, 4H) pre-activation
    const float* __restrict__ c_prev,  // (B, H)
    float* __restrict__ h_out, float* __restrict__ c_out,
    float* __restrict__ acts,          // (B, 4H) post-activation save
    float* __restrict__ tanhc,         // (B, H) save
    int64_t BH, int H) {
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; idx < BH; idx += stride) {
    int64_t b = idx / H;
    int64_t j = idx - b * H;
    const float* g4 = gates + b * 4 * H;
    float i = 1.0f / (1.0f + __expf(-g4[j]));
    float f = 1.0f / (1.0f + __expf(-g4[H + j]));
    float g = tanhf(g4[2 * H + j]);
    float o = 1.0f / (1.0f + __expf(-g4[3 * H + j]));
    float c = f * c_prev[idx] + i * g;
    float tc = tanhf(c);
    h_out[idx] = o * tc;
    c_out[idx] = c;
    float* a4 = acts + b * 4 * H;
    a4[j] = i;
    a4[H + j] = f;
    a4[2 * H + j] = g;
    a4[3 * H + j] = o;
    tanhc[idx] = tc;
  }
}

__global__ void lstm_cell_bwd_kernel(
    const float* __restrict__ dh,      // (B, H) recurrent dL/dh_t
    const float* __restrict__ gout_t,  // (B, H) per-step output grad or null
    const float* __restrict__ dc_in,   // (B, H) dL/dc_t from t+1
    const float* __restrict__ acts,    // (B, 4H) saved post-activations
    const float* __restrict__ tanhc,   // (B, H)
    const float* __restrict__ c_prev,  // (B, H)
    float* __restrict__ dgates,        // (B, 4H) pre-activation grads
    float* __restrict__ dc_prev,       // (B, H)
    int64_t BH, int H) {
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; idx < BH; idx += stride) {
    int64_t b = idx / H;
    int64_t j = idx - b * H;
    const float* a4 = acts + b * 4 * H;
    float i = a4[j], f = a4[H + j], g = a4[2 * H + j], o = a4[3 * H + j];
    float tc = tanhc[idx];
    float dhv = dh[idx] + (gout_t ? gout_t[idx] : 0.0f);
    float do_ = dhv * tc;
    float dct = dc_in[idx] + dhv * o * (1.0f - tc * tc);
    float di = dct * g;
    float df = dct * c_prev[idx];
    float dg = dct * i;
    dc_prev[idx] = dct * f;
    float* d4 = dgates + b * 4 * H;
    d4[j] = di * i * (1.0f - i);
    d4[H + j] = df * f * (1.0f - f);
    d4[2 * H + j] = dg * (1.0f - g * g);
    d4[3 * H + j] = do_ * o * (1.0f - o);
  }
}
}  // namespace

void lstm_cell_fwd(torch::Tensor gates, torch::Tensor c_prev,
                   torch::Tensor h_out, torch::Tensor c_out,
                   torch::Tensor acts, torch::Tensor tanhc) {
  int64_t B = c_prev.size(0), H = c_prev.size(1);
  hipLaunchKernelGGL(lstm_cell_fwd_kernel, dim3(grid_for(B * H)), dim3(kBlock),
                     0, cur_stream(), gates.data_ptr<float>(),
                     c_prev.data_ptr<float>(), h_out.data_ptr<float>(),
                     c_out.data_ptr<float>(), acts.data_ptr<float>(),
                     tanhc.data_ptr<float>(), B * H, (int)H);
}

void lstm_cell_bwd(torch::Tensor dh, torch::Tensor gout_t, torch::Tensor dc_in,
                   torch::Tensor acts, torch::Tensor tanhc,
                   torch::Tensor c_prev, torch::Tensor dgates,
                   torch::Tensor dc_prev) {
  int64_t B = dh.size(0), H = dh.size(1);
  const float* gp = gout_t.numel() ? gout_t.data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(lstm_cell_bwd_kernel, dim3(grid_for(B * H)), dim3(kBlock),
                     0, cur_stream(), dh.data_ptr<float>(), gp,
                     dc_in.data_ptr<float>(), acts.data_ptr<float>(),
                     tanhc.data_ptr<float>(), c_prev.data_ptr<float>(),
                     dgates.data_ptr<float>(), dc_prev.data_ptr<float>(),
                     B * H, (int)H);
}

// K9: fused IMPALA policy objective (IMPALA/Learner.py:95-114):
//   obj = mean_i( log pi(a_i) * adv_i ) + er * mean_i( H_i )
// forward computes softmax stats in one pass (A is small); backward emits
// d(-obj)/dlogits in closed form, so the log_softmax/exp/gather/entropy
// chain (and its autograd tape) collapses into 2 kernels.
namespace {
__global__ void policy_loss_fwd_kernel(
    const float* __restrict__ logits,  // (N, A)
    const int64_t* __restrict__ act, const float* __restrict__ adv,
    int N, int A, float er,
    float* __restrict__ obj_out,      // scalar, pre-zeroed (atomicAdd)
    float* __restrict__ ent_out,      // scalar, pre-zeroed
    float* __restrict__ logpa_out,    // (N,)
    float* __restrict__ pi_save,      // (N, A)
    float* __restrict__ H_save) {     // (N,)
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  float obj = 0.0f, entv = 0.0f;
  if (i < N) {
    const float* row = logits + (int64_t)i * A;
    float mx = row[0];
    for (int j = 1; j < A; ++j) mx = fmaxf(mx, row[j]);
    float Z = 0.0f;
    for (int j = 0; j < A; ++j) Z += __expf(row[j] - mx);
    float logZ = __logf(Z) + mx;
    float H = 0.0f;
    float* pr = pi_save + (int64_t)i * A;
    for (int j = 0; j < A; ++j) {
      float lp = row[j] - logZ;
      float p = __expf(lp);
      pr[j] = p;
      H -= p * lp;
    }
    H_save[i] = H;
    float lpa = row[act[i]] - logZ;
    logpa_out[i] = lpa;
    float invN = 1.0f / N;
    obj = lpa * adv[i] * invN;
    entv = H * invN;
  }
  for (int off = 32; off > 0; off >>= 1) {
    obj += __shfl_down(obj, off, 64);
    entv += __shfl_down(entv, off, 64);
  }
  if ((threadIdx.x & 63) == 0) {
    if (obj != 0.0f) atomicAdd(obj_out, obj);
    if (entv != 0.0f) atomicAdd(ent_out, entv);
  }
}

__global__ void policy_loss_bwd_kernel(
    const float* __restrict__ pi_save, const float* __restrict__ H_save,
    const int64_t* __restrict__ act, const float* __restrict__ adv,
    const float* __restrict__ gout,  // upstream grad scalar
    int N, int A, float er, float gsign, float* __restrict__ dlogits) {
  int64_t k = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (k >= (int64_t)N * A) return;
  int i = (int)(k / A);
  int j = (int)(k - (int64_t)i * A);
  float p = pi_save[k];
  float lp = __logf(fmaxf(p, 1e-30f));
  float invN = 1.0f / N;
  // d obj / d logit_ij = [ adv_i*(1{j=a} - p) - er * p*(lp + H_i) ] / N;
  // gsign = +1 when the upstream node is obj itself, -1 when it is
  // loss = -obj + critic (the fused-loss path).
  float d = adv[i] * (((int)act[i] == j ? 1.0f : 0.0f) - p)
            - er * p * (lp + H_save[i]);
  dlogits[k] = gsign * gout[0] * d * invN;
}

// Fused IMPALA total loss: loss = -(mean(logpa*adv) + er*mean_H) +
// 0.5*mean((v - vs)^2). mean_H arrives pre-reduced (policy_softmax_stats
// already accumulated it). ONE block — n = B*T is a few thousand — with
// plain stores at the end (no pre-zero fill, no atomics). Replaces the
// ~10-launch torch composition (mul, 3 means, mse, neg, add, broadcasts).
__global__ void impala_loss_fwd_kernel(
    const float* __restrict__ logpa, const float* __restrict__ adv,
    const float* __restrict__ mean_H, const float* __restrict__ v,
    const float* __restrict__ vs, int64_t n, float er,
    float* __restrict__ loss_out, float* __restrict__ obj_out,
    float* __restrict__ critic_out) {
  float s1 = 0.0f, s3 = 0.0f;
  for (int64_t i = threadIdx.x; i < n; i += blockDim.x) {
    s1 += logpa[i] * adv[i];
    float d = v[i] - vs[i];
    s3 += d * d;
  }
  for (int off = 32; off > 0; off >>= 1) {
    s1 += __shfl_down(s1, off, 64);
    s3 += __shfl_down(s3, off, 64);
  }
  __shared__ float w1[kBlock / 64], w3[kBlock / 64];
  if ((threadIdx.x & 63) == 0) {
    w1[threadIdx.x >> 6] = s1;
    w3[threadIdx.x >> 6] = s3;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float a = 0.0f, c = 0.0f;
    for (int w = 0; w < kBlock / 64; ++w) { a += w1[w]; c += w3[w]; }
    float invn = 1.0f / (float)n;
    float obj = a * invn + er * mean_H[0];
    float critic = 0.5f * c * invn;
    loss_out[0] = -obj + critic;
    obj_out[0] = obj;
    critic_out[0] = critic;
  }
}

// Whole-output IMPALA backward: writes d loss / d out for the raw network
// head out = (B*(T+1), A+1) in ONE launch — policy grad into columns 0..A-1
// for t<T, critic grad into column A for t<T, zeros elsewhere (t==T rows
// feed only the detached bootstrap). Replaces two SliceBackward
// zeros+copy+accumulate chains in autograd.
__global__ void impala_out_bwd_kernel(
    const float* __restrict__ pi_save,  // (B*T, A)
    const float* __restrict__ H_save,   // (B*T,)
    const int64_t* __restrict__ act,    // (B*T,)
    const float* __restrict__ adv,      // (B*T,)
    const float* __restrict__ v,        // (B*T,)
    const float* __restrict__ vs,       // (B*T,)
    const float* __restrict__ gloss, int B, int T, int A, float er,
    float* __restrict__ dout) {
  const int64_t total = (int64_t)B * (T + 1) * (A + 1);
  int64_t k = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float n_inv = 1.0f / (float)((int64_t)B * T);
  for (; k < total; k += stride) {
    int64_t r = k / (A + 1);
    int j = (int)(k - r * (A + 1));
    int b = (int)(r / (T + 1));
    int t = (int)(r - (int64_t)b * (T + 1));
    float g = 0.0f;
    if (t < T) {
      int64_t i = (int64_t)b * T + t;
      if (j < A) {
        float p = pi_save[i * A + j];
        float lp = __logf(fmaxf(p, 1e-30f));
        float d = adv[i] * (((int)act[i] == j ? 1.0f : 0.0f) - p)
                  - er * p * (lp + H_save[i]);
        g = -gloss[0] * d * n_inv;  // loss = -obj + critic
      } else {
        g = gloss[0] * (v[i] - vs[i]) * n_inv;
      }
    }
    dout[k] = g;
  }
}
}  // namespace

void policy_loss_fwd(torch::Tensor logits, torch::Tensor act, torch::Tensor adv,
                     double er, torch::Tensor obj_out, torch::Tensor ent_out,
                     torch::Tensor logpa_out, torch::Tensor pi_save,
                     torch::Tensor H_save) {
  int N = (int)logits.size(0), A = (int)logits.size(1);
  hipLaunchKernelGGL(policy_loss_fwd_kernel, dim3(ceil_div(N, kBlock)),
                     dim3(kBlock), 0, cur_stream(), logits.data_ptr<float>(),
                     act.data_ptr<int64_t>(), adv.data_ptr<float>(), N, A,
                     (float)er, obj_out.data_ptr<float>(),
                     ent_out.data_ptr<float>(), logpa_out.data_ptr<float>(),
                     pi_save.data_ptr<float>(), H_save.data_ptr<float>());
}

void policy_loss_bwd(torch::Tensor pi_save, torch::Tensor H_save,
                     torch::Tensor act, torch::Tensor adv, torch::Tensor gout,
                     double er, double gsign, torch::Tensor dlogits) {
  int N = (int)pi_save.size(0), A = (int)pi_save.size(1);
  hipLaunchKernelGGL(policy_loss_bwd_kernel,
                     dim3(ceil_div((int64_t)N * A, kBlock)), dim3(kBlock), 0,
                     cur_stream(), pi_save.data_ptr<float>(),
                     H_save.data_ptr<float>(), act.data_ptr<int64_t>(),
                     adv.data_ptr<float>(), gout.data_ptr<float>(), N, A,
                     (float)er, (float)gsign, dlogits.data_ptr<float>());
}

void impala_loss_fwd(torch::Tensor logpa, torch::Tensor adv,
                     torch::Tensor mean_H, torch::Tensor v, torch::Tensor vs,
                     double er, torch::Tensor loss_out, torch::Tensor obj_out,
                     torch::Tensor critic_out) {
  int64_t n = logpa.numel();
  TORCH_CHECK(adv.numel() == n && v.numel() == n && vs.numel() == n,
              "impala_loss_fwd length mismatch");
  for (auto* t : {&logpa, &adv, &v, &vs}) DRL_CHECK_CONTIG((*t));
  hipLaunchKernelGGL(impala_loss_fwd_kernel, dim3(1), dim3(kBlock), 0,
                     cur_stream(), logpa.data_ptr<float>(),
                     adv.data_ptr<float>(), mean_H.data_ptr<float>(),
                     v.data_ptr<float>(), vs.data_ptr<float>(), n, (float)er,
                     loss_out.data_ptr<float>(), obj_out.data_ptr<float>(),
                     critic_out.data_ptr<float>());
}

void impala_out_bwd(torch::Tensor pi_save, torch::Tensor H_save,
                    torch::Tensor act, torch::Tensor adv, torch::Tensor v,
                    torch::Tensor vs, torch::Tensor gloss, int64_t B,
                    int64_t T, int64_t A, double er, torch::Tensor dout) {
  int64_t total = B * (T + 1) * (A + 1);
  TORCH_CHECK(dout.numel() == total && dout.is_contiguous(),
              "impala_out_bwd dout shape mismatch");
  TORCH_CHECK(pi_save.numel() == B * T * A && v.numel() == B * T &&
              vs.numel() == B * T && act.numel() == B * T);
  hipLaunchKernelGGL(impala_out_bwd_kernel, dim3(grid_for(total, 2)),
                     dim3(kBlock), 0, cur_stream(), pi_save.data_ptr<float>(),
                     H_save.data_ptr<float>(), act.data_ptr<int64_t>(),
                     adv.data_ptr<float>(), v.data_ptr<float>(),
                     vs.data_ptr<float>(), gloss.data_ptr<float>(), (int)B,
                     (int)T, (int)A, (float)er, dout.data_ptr<float>());
}

// K12: fused optimizers over the flat master buffers. torch's capturable
// foreach RMSprop costs ~170 us/step for a 1.7M-param model (6 x
// multi_tensor_apply passes, profiles/); the whole centered-RMSprop update
// is one memory-bound pass here. Math matches torch.optim exactly.
namespace {
__global__ void rmsprop_step_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ sq, float* __restrict__ ga, float* __restrict__ mom,
    int64_t n, float lr, float alpha, float eps, float wd, float mu,
    bool centered, bool has_mom) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float grad = g[i] + wd * p[i];
    float s = alpha * sq[i] + (1.0f - alpha) * grad * grad;
    sq[i] = s;
    float avg;
    if (centered) {
      float a = alpha * ga[i] + (1.0f - alpha) * grad;
      ga[i] = a;
      avg = sqrtf(fmaxf(s - a * a, 0.0f)) + eps;
    } else {
      avg = sqrtf(s) + eps;
    }
    float upd = grad / avg;
    if (has_mom) {
      float m = mu * mom[i] + upd;
      mom[i] = m;
      upd = m;
    }
    p[i] -= lr * upd;
  }
}

__global__ void adam_bump_kernel(float* step) {
  if (threadIdx.x == 0 && blockIdx.x == 0) step[0] += 1.0f;
}

__global__ void adam_step_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    const float* __restrict__ step, int64_t n, float lr, float b1, float b2,
    float eps, float wd) {
  float t = step[0];
  float bc1 = 1.0f - __powf(b1, t);
  float bc2 = 1.0f - __powf(b2, t);
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float grad = g[i] + wd * p[i];
    float mi = b1 * m[i] + (1.0f - b1) * grad;
    float vi = b2 * v[i] + (1.0f - b2) * grad * grad;
    m[i] = mi;
    v[i] = vi;
    p[i] -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
  }
}
}  // namespace

void rmsprop_step(torch::Tensor p, torch::Tensor g, torch::Tensor sq,
                  torch::Tensor ga, torch::Tensor mom, double lr, double alpha,
                  double eps, double wd, double mu, bool centered,
                  bool has_mom) {
  int64_t n = p.numel();
  hipLaunchKernelGGL(rmsprop_step_kernel, dim3(grid_for(n, 4)), dim3(kBlock), 0,
                     cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
                     sq.data_ptr<float>(),
                     centered ? ga.data_ptr<float>() : sq.data_ptr<float>(),
                     has_mom ? mom.data_ptr<float>() : sq.data_ptr<float>(),
                     n, (float)lr, (float)alpha, (float)eps, (float)wd,
                     (float)mu, centered, has_mom);
}

void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, torch::Tensor step, double lr, double b1,
               double b2, double eps, double wd) {
  int64_t n = p.numel();
  hipLaunchKernelGGL(adam_bump_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     step.data_ptr<float>());
  hipLaunchKernelGGL(adam_step_kernel, dim3(grid_for(n, 4)), dim3(kBlock), 0,
                     cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     step.data_ptr<float>(), n, (float)lr, (float)b1,
                     (float)b2, (float)eps, (float)wd);
}

// K5 (full step): fused LSTM timestep — gates = xp_t + h_in @ W_hh^T on
// fp32 MFMA (v_mfma_f32_16x16x4_f32, exact fp32) + the cell nonlinearity,
// ONE kernel per timestep (vs hipBLASLt addmm ~8 us + cell ~5 us).
//   Decomposition: grid = H/16 blocks; block = 4 waves; wave w computes the
//   16-column slice of GATE w (torch gate order i,f,g,o along 4H), so after
//   one barrier the block holds all 4 gates for its 16 h-columns and
//   finishes the cell in-block. h_in (B x H) is staged once in LDS.
//   Requires B <= 32 (R2D2 cfg batch); caller falls back otherwise.
namespace {
constexpr int kLstmMaxB = 32;
using f32x4 = __attribute__((ext_vector_type(4))) float;

template <int H>
__global__ __launch_bounds__(256) void lstm_step_fused_kernel(
    const float* __restrict__ xp_t,   // (B, 4H) hoisted input projection
    const float* __restrict__ h_in,   // (B, H)
    const float* __restrict__ c_in,   // (B, H)
    const float* __restrict__ w_hh,   // (4H, H) row-major (torch layout)
    float* __restrict__ h_out, float* __restrict__ c_out,
    float* __restrict__ acts,         // (B, 4H)
    float* __restrict__ tanhc,        // (B, H)
    int B) {
  constexpr int PAD = 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* h_lds = reinterpret_cast<float*>(smem);           // [32][H+PAD]
  float* gbuf = h_lds + 32 * (H + PAD);                    // [4][32][16]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;   // = gate index
  const int s16 = blockIdx.x * 16;  // h-column slice base

  // ---- stage h_in (zero-padded rows) ----
  for (int i = tid * 4; i < 32 * H; i += 256 * 4) {
    int row = i / H;
    int k = i - row * H;
    float4 v = {0.f, 0.f, 0.f, 0.f};
    if (row < B) v = *reinterpret_cast<const float4*>(h_in + row * H + k);
    *reinterpret_cast<float4*>(h_lds + row * (H + PAD) + k) = v;
  }
  __syncthreads();

  // ---- MFMA: 2 row-fragments x 16 cols, K = H in 16-wide groups ----
  const int col = wave * H + s16 + (lane & 15);  // gate column in 4H
  f32x4 acc[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
  const int ksub = (lane >> 4) * 4;
  const float* wrow = w_hh + (int64_t)col * H;
#pragma unroll 4
  for (int kb = 0; kb < H; kb += 16) {
    float4 b4 = *reinterpret_cast<const float4*>(wrow + kb + ksub);
    float4 a0 = *reinterpret_cast<const float4*>(
        h_lds + (lane & 15) * (H + PAD) + kb + ksub);
    float4 a1 = *reinterpret_cast<const float4*>(
        h_lds + (16 + (lane & 15)) * (H + PAD) + kb + ksub);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0[j], b4[j], acc[0], 0, 0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1[j], b4[j], acc[1], 0, 0, 0);
    }
  }
  // ---- add xp, park gates in LDS ----
  const int crow = (lane >> 4) * 4;
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = rf * 16 + crow + r;
      int rclamp = row < B ? row : 0;
      float v = acc[rf][r] + xp_t[(int64_t)rclamp * 4 * H + col];
      gbuf[(wave * 32 + row) * 16 + (lane & 15)] = v;
    }
  __syncthreads();

  // ---- cell phase: 32 rows x 16 cols, 2 elems per thread ----
  for (int e = tid; e < 32 * 16; e += 256) {
    int row = e / 16;
    int hc = e - row * 16;
    if (row >= B) continue;
    float gi = gbuf[(0 * 32 + row) * 16 + hc];
    float gf = gbuf[(1 * 32 + row) * 16 + hc];
    float gg = gbuf[(2 * 32 + row) * 16 + hc];
    float go = gbuf[(3 * 32 + row) * 16 + hc];
    float i_ = 1.0f / (1.0f + __expf(-gi));
    float f_ = 1.0f / (1.0f + __expf(-gf));
    float g_ = tanhf(gg);
    float o_ = 1.0f / (1.0f + __expf(-go));
    int64_t hidx = (int64_t)row * H + s16 + hc;
    float c = f_ * c_in[hidx] + i_ * g_;
    float tc = tanhf(c);
    h_out[hidx] = o_ * tc;
    c_out[hidx] = c;
    tanhc[hidx] = tc;
    float* a4 = acts + (int64_t)row * 4 * H + s16 + hc;
    a4[0] = i_;
    a4[H] = f_;
    a4[2 * H] = g_;
    a4[3 * H] = o_;
  }
}
}  // namespace

// returns false if the geometry is unsupported (caller uses the 2-kernel path)
bool lstm_step_fused(torch::Tensor xp_t, torch::Tensor h_in, torch::Tensor c_in,
                     torch::Tensor w_hh, torch::Tensor h_out,
                     torch::Tensor c_out, torch::Tensor acts,
                     torch::Tensor tanhc) {
  int B = (int)h_in.size(0), H = (int)h_in.size(1);
  if (B > kLstmMaxB || H != 512) return false;
  constexpr int HH = 512;
  int lds = (32 * (HH + 4) + 4 * 32 * 16) * sizeof(float);
  hipLaunchKernelGGL(lstm_step_fused_kernel<HH>, dim3(HH / 16), dim3(256), lds,
                     cur_stream(), xp_t.data_ptr<float>(),
                     h_in.data_ptr<float>(), c_in.data_ptr<float>(),
                     w_hh.data_ptr<float>(), h_out.data_ptr<float>(),
                     c_out.data_ptr<float>(), acts.data_ptr<float>(),
                     tanhc.data_ptr<float>(), B);
  return true;
}

// K5 (persistent): the whole T-step recurrence in ONE launch. The graphed
// per-step loop is kernel-latency-floor-bound (~4-5 us/kernel x 2 kernels x
// T steps, profiles/); here the 32 blocks stay resident and synchronize
// with an agent-scope release/acquire grid barrier between timesteps
// (guide §6 G16: every storing wave drains vmcnt, one lane releases +
// arrives on a monotonic counter, consumers poll relaxed then acquire).
// Grid = H/16 = 32 blocks — trivially co-resident on 256 CUs; spins are
// bounded so a mis-launch exits instead of wedging the GPU.
namespace {

#define DRL_RLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT
typedef unsigned int __attribute__((address_space(1))) gu32_t;

template <int H>
__global__ __launch_bounds__(256) void lstm_seq_persistent_kernel(
    const float* __restrict__ xp,     // (T, B, 4H)
    float* __restrict__ hs,           // (T+1, B, H); hs[0] pre-filled
    float* __restrict__ cs,           // (T+1, B, H); cs[0] pre-filled
    const float* __restrict__ w_hh,   // (4H, H)
    float* __restrict__ acts,         // (T, B, 4H)
    float* __restrict__ tanhc,        // (T, B, H)
    unsigned int* __restrict__ ctr,   // zeroed before launch
    int B, int T) {
  constexpr int PAD = 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* h_lds = reinterpret_cast<float*>(smem);  // [32][H+PAD]
  float* gbuf = h_lds + 32 * (H + PAD);           // [4][32][16]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;  // gate index
  const int s16 = blockIdx.x * 16;
  const int col = wave * H + s16 + (lane & 15);
  const float* wrow = w_hh + (int64_t)col * H;
  const int ksub = (lane >> 4) * 4;
  const unsigned nblocks = gridDim.x;

  for (int t = 0; t < T; ++t) {
    const float* h_in = hs + (int64_t)t * B * H;
    const float* c_in = cs + (int64_t)t * B * H;
    const float* xp_t = xp + (int64_t)t * B * 4 * H;
    // ---- stage h (zero-padded rows) ----
    for (int i = tid * 4; i < 32 * H; i += 256 * 4) {
      int row = i / H;
      int k = i - row * H;
      float4 v = {0.f, 0.f, 0.f, 0.f};
      if (row < B) v = *reinterpret_cast<const float4*>(h_in + row * H + k);
      *reinterpret_cast<float4*>(h_lds + row * (H + PAD) + k) = v;
    }
    __syncthreads();
    // ---- hh GEMM on fp32 MFMA ----
    f32x4 acc[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
#pragma unroll 4
    for (int kb = 0; kb < H; kb += 16) {
      float4 b4 = *reinterpret_cast<const float4*>(wrow + kb + ksub);
      float4 a0 = *reinterpret_cast<const float4*>(
          h_lds + (lane & 15) * (H + PAD) + kb + ksub);
      float4 a1 = *reinterpret_cast<const float4*>(
          h_lds + (16 + (lane & 15)) * (H + PAD) + kb + ksub);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        acc[0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0[j], b4[j], acc[0], 0, 0, 0);
        acc[1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1[j], b4[j], acc[1], 0, 0, 0);
      }
    }
    const int crow = (lane >> 4) * 4;
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = rf * 16 + crow + r;
        int rclamp = row < B ? row : 0;
        gbuf[(wave * 32 + row) * 16 + (lane & 15)] =
            acc[rf][r] + xp_t[(int64_t)rclamp * 4 * H + col];
      }
    __syncthreads();
    // ---- cell + publish h/c for step t+1 ----
    float* h_out = hs + (int64_t)(t + 1) * B * H;
    float* c_out = cs + (int64_t)(t + 1) * B * H;
    for (int e = tid; e < 32 * 16; e += 256) {
      int row = e / 16;
      int hc = e - row * 16;
      if (row >= B) continue;
      float gi = gbuf[(0 * 32 + row) * 16 + hc];
      float gf = gbuf[(1 * 32 + row) * 16 + hc];
      float gg = gbuf[(2 * 32 + row) * 16 + hc];
      float go = gbuf[(3 * 32 + row) * 16 + hc];
      float i_ = 1.0f / (1.0f + __expf(-gi));
      float f_ = 1.0f / (1.0f + __expf(-gf));
      float g_ = tanhf(gg);
      float o_ = 1.0f / (1.0f + __expf(-go));
      int64_t hidx = (int64_t)row * H + s16 + hc;
      float c = f_ * c_in[hidx] + i_ * g_;
      float tc = tanhf(c);
      h_out[hidx] = o_ * tc;
      c_out[hidx] = c;
      tanhc[(int64_t)t * B * H + hidx] = tc;
      float* a4 = acts + (int64_t)t * B * 4 * H + (int64_t)row * 4 * H + s16 + hc;
      a4[0] = i_;
      a4[H] = f_;
      a4[2 * H] = g_;
      a4[3 * H] = o_;
    }
    // ---- grid barrier (G16 recipe): drain stores, release, arrive, poll,
    // acquire. Monotonic counter: target = nblocks * (t+1).
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // every storing wave
    __syncthreads();
    if (tid == 0) {
      __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // G16 pitfall 12
      __hip_atomic_fetch_add((gu32_t*)ctr, 1u, DRL_RLX_AGENT);
      const unsigned target = nblocks * (unsigned)(t + 1);
      unsigned spins = 0;
      while (__hip_atomic_load((gu32_t*)ctr, DRL_RLX_AGENT) < target) {
        __builtin_amdgcn_s_sleep(4);
        if (++spins > 5000000u) break;  // bounded: exit instead of wedging
      }
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    }
    __syncthreads();
  }
}
}  // namespace

bool lstm_seq_persistent(torch::Tensor xp, torch::Tensor hs, torch::Tensor cs,
                         torch::Tensor w_hh, torch::Tensor acts,
                         torch::Tensor tanhc, torch::Tensor ctr) {
  int T = (int)xp.size(0), B = (int)xp.size(1), H = (int)hs.size(2);
  if (B > kLstmMaxB || H != 512) return false;
  constexpr int HH = 512;
  int lds = (32 * (HH + 4) + 4 * 32 * 16) * sizeof(float);
  hipLaunchKernelGGL(lstm_seq_persistent_kernel<HH>, dim3(HH / 16), dim3(256),
                     lds, cur_stream(), xp.data_ptr<float>(),
                     hs.data_ptr<float>(), cs.data_ptr<float>(),
                     w_hh.data_ptr<float>(), acts.data_ptr<float>(),
                     tanhc.data_ptr<float>(),
                     (unsigned int*)ctr.data_ptr(), B, T);
  return true;
}

// ===========================================================================
// K5 v2 (round 2): ONE kernel per LSTM timestep, forward AND backward, with
// the hh GEMM on bf16 MFMA (v_mfma_f32_16x16x32_bf16, fp32 accumulate).
//
// Why: the round-1 per-step path is 2 kernels/step each direction
// (hipBLASLt addmm + fused cell = 12.8 us fwd, 18.1 us bwd measured) and is
// kernel-launch-floor-bound; the round-1 one-kernel fp32-MFMA step lost
// because f32 MFMA issues at 1/16 the bf16 rate (256 MFMA x 32 cyc on a
// 128-wave grid). bf16 operands cut the MFMA count 8x (K-step 32 vs 4), so
// one kernel per step runs at the launch floor:
//   fwd:  gates = xp_t + h_t @ W_hh^T  ->  cell  (h kept as a bf16 shadow)
//   bwd:  dh_t = dgates_{t+1} @ W_hh   ->  cell-bwd -> dgates_t (fp32+bf16)
// The backward fusion is only possible because the K=2048 GEMM fits one
// block's wave budget in bf16. Gates/cell state/saves stay fp32; only the
// GEMM operands (h, dgates, W) are bf16-rounded — the same precision class
// as the bf16 conv trunk feeding the same fp32 master (R2D2/Learner.py:94-121
// is the replaced op chain).
//
// MFMA fragment maps (guide §3, same as conv_mfma.hip):
//   A: lane l holds A[row=l&15][k=(l>>4)*8+j], j=0..7
//   B: lane l holds B[k=(l>>4)*8+j][col=l&15]
//   C/D: lane l holds C[row=(l>>4)*4+r][col=l&15], r=0..3
// ===========================================================================
namespace {

using bf16x8_k5 = __attribute__((ext_vector_type(8))) __bf16;

__device__ __forceinline__ float sigm_f(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// fwd: grid = H/16 blocks, 4 waves = 4 gates (torch order i,f,g,o), each
// wave computes a (32 x 16) C tile over K=H.
template <int H>
__global__ __launch_bounds__(256) void lstm_step_fwd_bf16_kernel(
    const float* __restrict__ xp_t,    // (B, 4H)
    const __bf16* __restrict__ h_bf,   // (B, H) bf16 shadow of h_t
    const float* __restrict__ c_in,    // (B, H)
    const __bf16* __restrict__ w_bf,   // (4H, H) bf16 copy of W_hh
    float* __restrict__ h_out,         // (B, H)
    float* __restrict__ c_out,         // (B, H)
    __bf16* __restrict__ h_bf_out,     // (B, H)
    float* __restrict__ acts,          // (B, 4H)
    float* __restrict__ tanhc,         // (B, H)
    int B) {
  __shared__ float gbuf[4][32][16];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;              // gate index
  const int s16 = blockIdx.x * 16;        // h-column slice base
  const int col = wave * H + s16 + (lane & 15);
  const __bf16* wrow = w_bf + (int64_t)col * H;
  const int kbase = (lane >> 4) * 8;
  const int r0 = lane & 15, r1 = 16 + (lane & 15);
  const int r0c = r0 < B ? r0 : 0, r1c = r1 < B ? r1 : 0;
  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll 4
  for (int kb = 0; kb < H; kb += 32) {
    bf16x8_k5 bfrag = *reinterpret_cast<const bf16x8_k5*>(wrow + kb + kbase);
    bf16x8_k5 a0 = *reinterpret_cast<const bf16x8_k5*>(
        h_bf + (int64_t)r0c * H + kb + kbase);
    bf16x8_k5 a1 = *reinterpret_cast<const bf16x8_k5*>(
        h_bf + (int64_t)r1c * H + kb + kbase);
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bfrag, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, bfrag, acc1, 0, 0, 0);
  }
  const int crow = (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row0 = crow + r, row1 = 16 + crow + r;
    int row0c = row0 < B ? row0 : 0, row1c = row1 < B ? row1 : 0;
    gbuf[wave][row0][lane & 15] =
        acc0[r] + xp_t[(int64_t)row0c * 4 * H + col];
    gbuf[wave][row1][lane & 15] =
        acc1[r] + xp_t[(int64_t)row1c * 4 * H + col];
  }
  __syncthreads();
  for (int e = tid; e < 32 * 16; e += 256) {
    int row = e >> 4;
    int hc = e & 15;
    if (row >= B) continue;
    float i_ = sigm_f(gbuf[0][row][hc]);
    float f_ = sigm_f(gbuf[1][row][hc]);
    float g_ = tanhf(gbuf[2][row][hc]);
    float o_ = sigm_f(gbuf[3][row][hc]);
    int64_t hidx = (int64_t)row * H + s16 + hc;
    float c = f_ * c_in[hidx] + i_ * g_;
    float tc = tanhf(c);
    float h = o_ * tc;
    h_out[hidx] = h;
    c_out[hidx] = c;
    h_bf_out[hidx] = (__bf16)h;
    tanhc[hidx] = tc;
    float* a4 = acts + (int64_t)row * 4 * H + s16 + hc;
    a4[0] = i_;
    a4[H] = f_;
    a4[2 * H] = g_;
    a4[3 * H] = o_;
  }
}

// bwd: grid = H/16 blocks; the 4 waves split K=4H four ways for the
// dh = dgates_{t+1} @ W_hh GEMM (partials reduced through LDS), then the
// block finishes cell-bwd for its 16 h-columns. At t=T-1 (dg_prev null)
// dh comes from dh_init instead.
template <int H>
__global__ __launch_bounds__(256) void lstm_step_bwd_bf16_kernel(
    const __bf16* __restrict__ dg_prev,  // (B, 4H) bf16 or null
    const float* __restrict__ dh_init,   // (B, H), used when dg_prev null
    const float* __restrict__ gout_t,    // (B, H) or null
    const float* __restrict__ dc_in,     // (B, H)
    const __bf16* __restrict__ w_t_bf,   // (H, 4H) bf16 = W_hh^T
    const float* __restrict__ acts,      // (B, 4H) saved at t
    const float* __restrict__ tanhc,     // (B, H)
    const float* __restrict__ c_prev,    // (B, H)
    float* __restrict__ dgates,          // (B, 4H) fp32 out
    __bf16* __restrict__ dgates_bf,      // (B, 4H) bf16 out
    float* __restrict__ dc_out,          // (B, H)
    int B) {
  __shared__ float partial[4][32][16];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int n0 = blockIdx.x * 16;  // h-column slice base
  if (dg_prev != nullptr) {
    const int kbase = (lane >> 4) * 8;
    const int r0 = lane & 15, r1 = 16 + (lane & 15);
    const int r0c = r0 < B ? r0 : 0, r1c = r1 < B ? r1 : 0;
    const __bf16* wcol = w_t_bf + (int64_t)(n0 + (lane & 15)) * 4 * H;
    f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = {0.f, 0.f, 0.f, 0.f};
    const int k0 = wave * H;  // this wave's K chunk: [k0, k0+H)
#pragma unroll 4
    for (int kb = 0; kb < H; kb += 32) {
      bf16x8_k5 bfrag =
          *reinterpret_cast<const bf16x8_k5*>(wcol + k0 + kb + kbase);
      bf16x8_k5 a0 = *reinterpret_cast<const bf16x8_k5*>(
          dg_prev + (int64_t)r0c * 4 * H + k0 + kb + kbase);
      bf16x8_k5 a1 = *reinterpret_cast<const bf16x8_k5*>(
          dg_prev + (int64_t)r1c * 4 * H + k0 + kb + kbase);
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bfrag, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, bfrag, acc1, 0, 0, 0);
    }
    const int crow = (lane >> 4) * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      partial[wave][crow + r][lane & 15] = acc0[r];
      partial[wave][16 + crow + r][lane & 15] = acc1[r];
    }
  }
  __syncthreads();
  for (int e = tid; e < 32 * 16; e += 256) {
    int row = e >> 4;
    int hc = e & 15;
    if (row >= B) continue;
    int64_t hidx = (int64_t)row * H + n0 + hc;
    float dhv;
    if (dg_prev != nullptr) {
      dhv = partial[0][row][hc] + partial[1][row][hc] + partial[2][row][hc] +
            partial[3][row][hc];
    } else {
      dhv = dh_init[hidx];
    }
    if (gout_t != nullptr) dhv += gout_t[hidx];
    const float* a4 = acts + (int64_t)row * 4 * H + n0 + hc;
    float i = a4[0], f = a4[H], g = a4[2 * H], o = a4[3 * H];
    float tc = tanhc[hidx];
    float do_ = dhv * tc;
    float dct = dc_in[hidx] + dhv * o * (1.0f - tc * tc);
    float di = dct * g;
    float df = dct * c_prev[hidx];
    float dg = dct * i;
    dc_out[hidx] = dct * f;
    float v0 = di * i * (1.0f - i);
    float v1 = df * f * (1.0f - f);
    float v2 = dg * (1.0f - g * g);
    float v3 = do_ * o * (1.0f - o);
    float* d4 = dgates + (int64_t)row * 4 * H + n0 + hc;
    d4[0] = v0;
    d4[H] = v1;
    d4[2 * H] = v2;
    d4[3 * H] = v3;
    __bf16* b4 = dgates_bf + (int64_t)row * 4 * H + n0 + hc;
    b4[0] = (__bf16)v0;
    b4[H] = (__bf16)v1;
    b4[2 * H] = (__bf16)v2;
    b4[3 * H] = (__bf16)v3;
  }
}
}  // namespace

bool lstm_step_fwd_bf16(torch::Tensor xp_t, torch::Tensor h_bf,
                        torch::Tensor c_in, torch::Tensor w_bf,
                        torch::Tensor h_out, torch::Tensor c_out,
                        torch::Tensor h_bf_out, torch::Tensor acts_t,
                        torch::Tensor tanhc_t) {
  int B = (int)c_in.size(0), H = (int)c_in.size(1);
  if (B > kLstmMaxB || H != 512) return false;
  constexpr int HH = 512;
  hipLaunchKernelGGL(lstm_step_fwd_bf16_kernel<HH>, dim3(HH / 16), dim3(256),
                     0, cur_stream(), xp_t.data_ptr<float>(),
                     (const __bf16*)h_bf.data_ptr(), c_in.data_ptr<float>(),
                     (const __bf16*)w_bf.data_ptr(), h_out.data_ptr<float>(),
                     c_out.data_ptr<float>(), (__bf16*)h_bf_out.data_ptr(),
                     acts_t.data_ptr<float>(), tanhc_t.data_ptr<float>(), B);
  return true;
}

bool lstm_step_bwd_bf16(torch::Tensor dg_prev, torch::Tensor dh_init,
                        torch::Tensor gout_t, torch::Tensor dc_in,
                        torch::Tensor w_t_bf, torch::Tensor acts_t,
                        torch::Tensor tanhc_t, torch::Tensor c_prev,
                        torch::Tensor dgates_t, torch::Tensor dg_bf_t,
                        torch::Tensor dc_out) {
  int B = (int)dc_in.size(0), H = (int)dc_in.size(1);
  if (B > kLstmMaxB || H != 512) return false;
  constexpr int HH = 512;
  const __bf16* dgp =
      dg_prev.numel() ? (const __bf16*)dg_prev.data_ptr() : nullptr;
  const float* gp = gout_t.numel() ? gout_t.data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(lstm_step_bwd_bf16_kernel<HH>, dim3(HH / 16), dim3(256),
                     0, cur_stream(), dgp, dh_init.data_ptr<float>(), gp,
                     dc_in.data_ptr<float>(),
                     (const __bf16*)w_t_bf.data_ptr(),
                     acts_t.data_ptr<float>(), tanhc_t.data_ptr<float>(),
                     c_prev.data_ptr<float>(), dgates_t.data_ptr<float>(),
                     (__bf16*)dg_bf_t.data_ptr(), dc_out.data_ptr<float>(), B);
  return true;
}

// ===========================================================================
// K3+K4 fused (round 2): whole-head dueling n-step double-DQN loss.
// The dueling epilogue Q = A - mean(A) + V of all THREE forwards (online s,
// online s', target s') is folded INTO the loss kernel (argmax over Q(s')
// equals argmax over A(s') since mean/V are row constants), and the
// backward writes the head gradients (g_adv, g_val) in closed form:
//   g_adv[i,j] = -coef_i * gout * (1{j==a_i} - 1/A),  g_val[i] = -coef_i*gout
// Replaces: 3x dueling_fwd + 6 bf16->f32 casts + dueling_bwd + dqn_loss_bwd
// (~10 launches -> 2). Reference op chains: cfg/ape_x.json:72-88 dueling
// graph + APE_X/Learner.py:83-114 loss.
// ===========================================================================
namespace {
template <typename T>
__global__ void dueling_dqn_loss_fwd_kernel(
    const T* __restrict__ adv_s, const T* __restrict__ val_s,
    const T* __restrict__ adv_on, const T* __restrict__ val_on,
    const T* __restrict__ adv_tg, const T* __restrict__ val_tg,
    const int64_t* __restrict__ act, const float* __restrict__ rew,
    const float* __restrict__ done, const float* __restrict__ w, int B, int A,
    float gamma_n, float alpha, float* __restrict__ loss_out /*pre-zeroed*/,
    float* __restrict__ prio_out, float* __restrict__ grad_coef,
    float* __restrict__ qmax_out /*pre-zeroed*/) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  float contrib = 0.0f, qm = 0.0f;
  if (i < B) {
    const int64_t ro = (int64_t)i * A;
    // a* = argmax_a Q_on(s',a) = argmax_a A_on(s',a)
    int a_star = 0;
    float best = drl_ld(adv_on + ro, 0);
    for (int a = 1; a < A; ++a) {
      float v = drl_ld(adv_on + ro, a);
      if (v > best) { best = v; a_star = a; }
    }
    float mean_tg = 0.0f;
    for (int a = 0; a < A; ++a) mean_tg += drl_ld(adv_tg + ro, a);
    mean_tg /= A;
    float q_tg = drl_ld(adv_tg + ro, a_star) - mean_tg + drl_ld(val_tg, i);
    float target = rew[i] + gamma_n * q_tg * (1.0f - done[i]);
    float mean_s = 0.0f, smax = drl_ld(adv_s + ro, 0);
    for (int a = 0; a < A; ++a) {
      float v = drl_ld(adv_s + ro, a);
      mean_s += v;
      smax = fmaxf(smax, v);
    }
    mean_s /= A;
    float vs = drl_ld(val_s, i);
    float q = drl_ld(adv_s + ro, (int)act[i]) - mean_s + vs;
    qm = (smax - mean_s + vs) / B;  // mean over rows of max_a Q(s,a)
    float raw = target - q;
    float td = fminf(1.0f, fmaxf(-1.0f, raw));
    prio_out[i] = __powf(fabsf(td) + 1e-7f, alpha);
    float in_range = (raw > -1.0f && raw < 1.0f) ? 1.0f : 0.0f;
    float invB = 1.0f / B;
    grad_coef[i] = w[i] * td * in_range * invB;  // dL/dQ(s,a) = -grad_coef
    contrib = 0.5f * w[i] * td * td * invB;
  }
  for (int off = 32; off > 0; off >>= 1) {
    contrib += __shfl_down(contrib, off, 64);
    qm += __shfl_down(qm, off, 64);
  }
  if ((threadIdx.x & 63) == 0) {
    if (contrib != 0.0f) atomicAdd(loss_out, contrib);
    if (qm != 0.0f) atomicAdd(qmax_out, qm);
  }
}

template <typename T>
__global__ void dueling_dqn_loss_bwd_kernel(
    const float* __restrict__ grad_coef, const int64_t* __restrict__ act,
    const float* __restrict__ gout, int B, int A, T* __restrict__ g_adv,
    T* __restrict__ g_val) {
  int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= B * A) return;
  int i = j / A;
  int a = j - i * A;
  float c = -grad_coef[i] * gout[0];
  float ind = (a == (int)act[i]) ? 1.0f : 0.0f;
  g_adv[j] = (T)(c * (ind - 1.0f / A));
  if (a == 0) g_val[i] = (T)c;
}
}  // namespace

void dueling_dqn_loss_fwd(torch::Tensor adv_s, torch::Tensor val_s,
                          torch::Tensor adv_on, torch::Tensor val_on,
                          torch::Tensor adv_tg, torch::Tensor val_tg,
                          torch::Tensor act, torch::Tensor rew,
                          torch::Tensor done, torch::Tensor w, double gamma_n,
                          double alpha, torch::Tensor loss_out,
                          torch::Tensor prio_out, torch::Tensor grad_coef,
                          torch::Tensor qmax_out) {
  int B = (int)adv_s.size(0), A = (int)adv_s.size(1);
  dim3 grid(ceil_div(B, kBlock)), blk(kBlock);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, adv_s.scalar_type(),
      "dueling_dqn_loss_fwd", [&] {
        hipLaunchKernelGGL(
            dueling_dqn_loss_fwd_kernel<scalar_t>, grid, blk, 0, cur_stream(),
            adv_s.data_ptr<scalar_t>(), val_s.data_ptr<scalar_t>(),
            adv_on.data_ptr<scalar_t>(), val_on.data_ptr<scalar_t>(),
            adv_tg.data_ptr<scalar_t>(), val_tg.data_ptr<scalar_t>(),
            act.data_ptr<int64_t>(), rew.data_ptr<float>(),
            done.data_ptr<float>(), w.data_ptr<float>(), B, A, (float)gamma_n,
            (float)alpha, loss_out.data_ptr<float>(),
            prio_out.data_ptr<float>(), grad_coef.data_ptr<float>(),
            qmax_out.data_ptr<float>());
      });
}

void dueling_dqn_loss_bwd(torch::Tensor grad_coef, torch::Tensor act,
                          torch::Tensor gout, torch::Tensor g_adv,
                          torch::Tensor g_val) {
  int B = (int)g_adv.size(0), A = (int)g_adv.size(1);
  dim3 grid(ceil_div((int64_t)B * A, kBlock)), blk(kBlock);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, g_adv.scalar_type(),
      "dueling_dqn_loss_bwd", [&] {
        hipLaunchKernelGGL(dueling_dqn_loss_bwd_kernel<scalar_t>, grid, blk, 0,
                           cur_stream(), grad_coef.data_ptr<float>(),
                           act.data_ptr<int64_t>(), gout.data_ptr<float>(), B,
                           A, g_adv.data_ptr<scalar_t>(),
                           g_val.data_ptr<scalar_t>());
      });
}

// ===========================================================================
// Host-side ingest pack (C2): AoS shm records -> SoA pinned staging in one
// GIL-free multithreaded pass. The Python per-field numpy copies held the
// GIL and starved the learner loop (measured 2.9 ms/step whole-node with
// the ingest thread vs 0.67 learner-only); this is one call per ring chunk.
// ===========================================================================
void pack_rows(torch::Tensor src, int64_t rec_size,
               std::vector<int64_t> offs, std::vector<int64_t> sizes,
               std::vector<torch::Tensor> dsts, int64_t dst_row) {
  TORCH_CHECK(src.device().is_cpu() && src.dtype() == torch::kUInt8);
  const uint8_t* s = src.data_ptr<uint8_t>();
  int64_t nrows = src.numel() / rec_size;
  int nf = (int)offs.size();
  std::vector<uint8_t*> dp(nf);
  std::vector<int64_t> stride(nf);
  for (int f = 0; f < nf; ++f) {
    dp[f] = (uint8_t*)dsts[f].data_ptr();
    stride[f] = dsts[f].numel() * dsts[f].element_size() / dsts[f].size(0);
    dp[f] += dst_row * stride[f];
    TORCH_CHECK(stride[f] == sizes[f], "dst row stride != field size");
  }
  at::parallel_for(0, nrows, 16, [&](int64_t b, int64_t e) {
    for (int64_t r = b; r < e; ++r) {
      const uint8_t* row = s + r * rec_size;
      for (int f = 0; f < nf; ++f)
        memcpy(dp[f] + r * sizes[f], row + offs[f], (size_t)sizes[f]);
    }
  });
}

// ===========================================================================
// K5 v3: persistent whole-sequence LSTM on bf16 MFMA — the entire T-step
// recurrence (fwd) / reversed scan (bwd) in ONE launch. The v2 per-step
// kernels measure 9.6 us fwd / 10.8 us bwd, dominated by launch+tail on a
// ~1-2 us workload; here 32 blocks stay resident and synchronize with the
// agent-scope release/acquire grid barrier between timesteps (G16 recipe,
// same as lstm_seq_persistent — which lost only because its fp32 MFMA
// inner loop was 8x the instruction count). Cell/carry state (c fwd,
// dc bwd) lives in REGISTERS: the thread->element map is fixed across
// timesteps, so only h (fwd) / dgates (bwd) cross block boundaries.
// ===========================================================================
namespace {

template <int H>
__global__ __launch_bounds__(256) void lstm_seq_fwd_bf16_kernel(
    const float* __restrict__ xp,      // (T, B, 4H)
    const float* __restrict__ c0,      // (B, H)
    const __bf16* __restrict__ w_bf,   // (4H, H)
    float* __restrict__ hs,            // (T+1, B, H); hs[0] ignored here
    float* __restrict__ cs,            // (T+1, B, H); cs[0] = c0 pre-filled
    __bf16* __restrict__ h_bfs,        // (T+1, B, H); [0] pre-filled
    float* __restrict__ acts,          // (T, B, 4H)
    float* __restrict__ tanhc,         // (T, B, H)
    unsigned int* __restrict__ ctr,    // zeroed before launch
    int B, int T) {
  // Latency discipline (the non-LDS version measured ~8-12 us/step: 16
  // serialized L2 round-trips per wave): the block's W slice (64 gate-rows
  // x H) is burst-staged into LDS ONCE and reused for all T steps; h is
  // burst-staged per step. Fragment reads then come from LDS.
  constexpr int LROW = H + 32;  // (LROW*2) % 256 == 64: spreads b128 quads
  __shared__ __bf16 wlds[64][LROW];
  __shared__ __bf16 hlds[32][LROW];
  __shared__ float gbuf[4][32][16];
  __shared__ float xplds[4][32][17];  // xp slice (scattered 64 B runs in
                                      // global; the per-lane scalar reads
                                      // in the park phase measured ~3 us)
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int s16 = blockIdx.x * 16;
  const int kbase = (lane >> 4) * 8;
  const unsigned nblocks = gridDim.x;

  // stage W: LDS row g*16+c <- w_bf[g*H + s16 + c][:]
  for (int base = tid * 8; base < 64 * H; base += 256 * 8) {
    const int r = base / H;
    const int k = base - r * H;
    const int g = r >> 4, c = r & 15;
    *reinterpret_cast<bf16x8_k5*>(&wlds[r][k]) =
        *reinterpret_cast<const bf16x8_k5*>(w_bf + ((int64_t)g * H + s16 + c) * H + k);
  }

  // register-resident cell state: thread e owns elements e, e+256 of the
  // block's (32 x 16) tile
  float c_reg[2];
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    int e = tid + u * 256;
    int row = e >> 4, hc = e & 15;
    c_reg[u] = (row < B) ? c0[(int64_t)row * H + s16 + hc] : 0.0f;
  }
  __syncthreads();

  for (int t = 0; t < T; ++t) {
    const __bf16* h_bf = h_bfs + (int64_t)t * B * H;
    const float* xp_t = xp + (int64_t)t * B * 4 * H;
    // burst-stage h_t (zero-pad rows >= B), 16 elems per thread round
    for (int base = tid * 16; base < 32 * H; base += 256 * 16) {
      const int r = base / H;
      const int k = base - r * H;
      bf16x8_k5 v0{}, v1{};
      if (r < B) {
        v0 = *reinterpret_cast<const bf16x8_k5*>(h_bf + (int64_t)r * H + k);
        v1 = *reinterpret_cast<const bf16x8_k5*>(h_bf + (int64_t)r * H + k + 8);
      }
      *reinterpret_cast<bf16x8_k5*>(&hlds[r][k]) = v0;
      *reinterpret_cast<bf16x8_k5*>(&hlds[r][k + 8]) = v1;
    }
    // burst-stage the xp slice: thread t covers half of (gate, row) run
    {
      const int run = tid >> 1;  // gate*32 + row
      const int g_ = run >> 5;
      const int row = run & 31;
      const int half = (tid & 1) * 8;
      const int rowc = row < B ? row : 0;
      const float* src =
          xp_t + (int64_t)rowc * 4 * H + g_ * H + s16 + half;
      float4 lo = *reinterpret_cast<const float4*>(src);
      float4 hi = *reinterpret_cast<const float4*>(src + 4);
      *reinterpret_cast<float4*>(&xplds[g_][row][half]) = lo;
      *reinterpret_cast<float4*>(&xplds[g_][row][half + 4]) = hi;
    }
    __syncthreads();
    f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll 4
    for (int kb = 0; kb < H; kb += 32) {
      bf16x8_k5 bfrag = *reinterpret_cast<const bf16x8_k5*>(
          &wlds[wave * 16 + (lane & 15)][kb + kbase]);
      bf16x8_k5 a0 = *reinterpret_cast<const bf16x8_k5*>(
          &hlds[lane & 15][kb + kbase]);
      bf16x8_k5 a1 = *reinterpret_cast<const bf16x8_k5*>(
          &hlds[16 + (lane & 15)][kb + kbase]);
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bfrag, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, bfrag, acc1, 0, 0, 0);
    }
    const int crow = (lane >> 4) * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row0 = crow + r, row1 = 16 + crow + r;
      gbuf[wave][row0][lane & 15] =
          acc0[r] + xplds[wave][row0][lane & 15];
      gbuf[wave][row1][lane & 15] =
          acc1[r] + xplds[wave][row1][lane & 15];
    }
    __syncthreads();
    float* h_out = hs + (int64_t)(t + 1) * B * H;
    float* c_out = cs + (int64_t)(t + 1) * B * H;
    __bf16* h_bf_out = h_bfs + (int64_t)(t + 1) * B * H;
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      int e = tid + u * 256;
      int row = e >> 4, hc = e & 15;
      if (row >= B) continue;
      float i_ = sigm_f(gbuf[0][row][hc]);
      float f_ = sigm_f(gbuf[1][row][hc]);
      float g_ = tanhf(gbuf[2][row][hc]);
      float o_ = sigm_f(gbuf[3][row][hc]);
      int64_t hidx = (int64_t)row * H + s16 + hc;
      float c = f_ * c_reg[u] + i_ * g_;
      c_reg[u] = c;
      float tc = tanhf(c);
      float h = o_ * tc;
      h_out[hidx] = h;
      c_out[hidx] = c;
      h_bf_out[hidx] = (__bf16)h;
      tanhc[(int64_t)t * B * H + hidx] = tc;
      // acts packed (T,B,H,4): ONE 16 B store instead of four 4 B stores
      // 2 KB apart — the barrier's vmcnt(0) drain paid for every scattered
      // store (layout is internal to the persistent fwd/bwd pair)
      *reinterpret_cast<float4*>(
          acts + (((int64_t)t * B + row) * H + s16 + hc) * 4) =
          float4{i_, f_, g_, o_};
    }
    // agent-scope grid barrier: h_bf_out must be visible to every block
    __syncthreads();
    if (tid == 0) {
      __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __hip_atomic_fetch_add((gu32_t*)ctr, 1u, DRL_RLX_AGENT);
      const unsigned target = nblocks * (unsigned)(t + 1);
      unsigned spins = 0;
      while (__hip_atomic_load((gu32_t*)ctr, DRL_RLX_AGENT) < target) {
        __builtin_amdgcn_s_sleep(4);
        if (++spins > 5000000u) break;  // bounded: exit instead of wedging
      }
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    }
    __syncthreads();
  }
}

template <int H>
__global__ __launch_bounds__(256) void lstm_seq_bwd_bf16_kernel(
    const float* __restrict__ dh_init,  // (B, H)
    const float* __restrict__ gout,     // (T, B, H)
    const float* __restrict__ dc_T,     // (B, H)
    const __bf16* __restrict__ w_t_bf,  // (H, 4H)
    const float* __restrict__ acts,     // (T, B, 4H)
    const float* __restrict__ tanhc,    // (T, B, H)
    const float* __restrict__ cs,       // (T+1, B, H)
    float* __restrict__ dgates,         // (T, B, 4H)
    __bf16* __restrict__ dgates_bf,     // (T, B, 4H)
    float* __restrict__ dc0_out,        // (B, H)
    unsigned int* __restrict__ ctr,     // zeroed before launch
    int B, int T) {
  // Latency discipline (see the fwd kernel): the block's W^T slice
  // (16 h-cols x 4H) is burst-staged into LDS once; dg_{t+1} (32 x 4H,
  // 128 KB bf16) is burst-staged per step in TWO 64 KB phases reusing one
  // buffer, with the k-split re-drawn so each wave covers a quarter of
  // each phase (the union over waves x phases is the full K; partials
  // still fold 4-ways through LDS).
  constexpr int LROWW = 4 * H + 32;
  constexpr int LROWD = 2 * H + 32;
  __shared__ __bf16 wtlds[16][LROWW];
  __shared__ __bf16 dglds[32][LROWD];
  __shared__ float partial[4][32][16];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int n0 = blockIdx.x * 16;
  const int kbase = (lane >> 4) * 8;
  const unsigned nblocks = gridDim.x;

  for (int base = tid * 8; base < 16 * 4 * H; base += 256 * 8) {
    const int c = base / (4 * H);
    const int k = base - c * 4 * H;
    *reinterpret_cast<bf16x8_k5*>(&wtlds[c][k]) =
        *reinterpret_cast<const bf16x8_k5*>(
            w_t_bf + (int64_t)(n0 + c) * 4 * H + k);
  }

  float dc_reg[2];
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    int e = tid + u * 256;
    int row = e >> 4, hc = e & 15;
    dc_reg[u] = (row < B) ? dc_T[(int64_t)row * H + n0 + hc] : 0.0f;
  }

  for (int it = 0; it < T; ++it) {
    const int t = T - 1 - it;
    if (it > 0) {
      const __bf16* dg_prev = dgates_bf + (int64_t)(t + 1) * B * 4 * H;
      f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ph = 0; ph < 2; ++ph) {
        __syncthreads();  // dglds reuse: prior phase reads must finish
        for (int base = tid * 16; base < 32 * 2 * H; base += 256 * 16) {
          const int r = base / (2 * H);
          const int k = base - r * 2 * H;
          bf16x8_k5 v0{}, v1{};
          if (r < B) {
            const __bf16* sp = dg_prev + (int64_t)r * 4 * H + ph * 2 * H + k;
            v0 = *reinterpret_cast<const bf16x8_k5*>(sp);
            v1 = *reinterpret_cast<const bf16x8_k5*>(sp + 8);
          }
          *reinterpret_cast<bf16x8_k5*>(&dglds[r][k]) = v0;
          *reinterpret_cast<bf16x8_k5*>(&dglds[r][k + 8]) = v1;
        }
        __syncthreads();
        const int kw0 = wave * (H / 2);  // this wave's quarter of the phase
#pragma unroll 4
        for (int kb = kw0; kb < kw0 + H / 2; kb += 32) {
          bf16x8_k5 bfrag = *reinterpret_cast<const bf16x8_k5*>(
              &wtlds[lane & 15][ph * 2 * H + kb + kbase]);
          bf16x8_k5 a0 = *reinterpret_cast<const bf16x8_k5*>(
              &dglds[lane & 15][kb + kbase]);
          bf16x8_k5 a1 = *reinterpret_cast<const bf16x8_k5*>(
              &dglds[16 + (lane & 15)][kb + kbase]);
          acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bfrag, acc0, 0, 0, 0);
          acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, bfrag, acc1, 0, 0, 0);
        }
      }
      const int crow = (lane >> 4) * 4;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        partial[wave][crow + r][lane & 15] = acc0[r];
        partial[wave][16 + crow + r][lane & 15] = acc1[r];
      }
    }
    __syncthreads();
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      int e = tid + u * 256;
      int row = e >> 4, hc = e & 15;
      if (row >= B) continue;
      int64_t hidx = (int64_t)row * H + n0 + hc;
      float dhv = (it > 0)
          ? partial[0][row][hc] + partial[1][row][hc] + partial[2][row][hc] +
                partial[3][row][hc]
          : dh_init[hidx];
      dhv += gout[(int64_t)t * B * H + hidx];
      const float4 a4 = *reinterpret_cast<const float4*>(
          acts + (((int64_t)t * B + row) * H + n0 + hc) * 4);
      float i = a4.x, f = a4.y, g = a4.z, o = a4.w;
      float tc = tanhc[(int64_t)t * B * H + hidx];
      float do_ = dhv * tc;
      float dct = dc_reg[u] + dhv * o * (1.0f - tc * tc);
      float di = dct * g;
      float df = dct * cs[(int64_t)t * B * H + hidx];
      float dg = dct * i;
      dc_reg[u] = dct * f;
      // park dgates in the (already-consumed) partial buffer: each thread
      // reads/writes only its own (row, hc) entries, so no cross-thread
      // hazard; a cooperative burst below emits coalesced 32 B runs
      // instead of 8 scalar stores 2 KB apart (barrier drain cost)
      partial[0][row][hc] = di * i * (1.0f - i);
      partial[1][row][hc] = df * f * (1.0f - f);
      partial[2][row][hc] = dg * (1.0f - g * g);
      partial[3][row][hc] = do_ * o * (1.0f - o);
    }
    __syncthreads();
    {
      // burst dgates (fp32, gate-major) + dgates_bf (bf16) from LDS:
      // thread pair covers one (gate, row): 16 contiguous cols
      const int pr = tid >> 1;        // 0..127 = gate*32 + row
      const int half = (tid & 1) * 8;
      const int g_ = pr >> 5;
      const int row = pr & 31;
      if (row < B) {
        const int64_t base =
            (int64_t)t * B * 4 * H + (int64_t)row * 4 * H + g_ * H + n0;
        float4 lo = *reinterpret_cast<const float4*>(&partial[g_][row][half]);
        float4 hi =
            *reinterpret_cast<const float4*>(&partial[g_][row][half + 4]);
        *reinterpret_cast<float4*>(dgates + base + half) = lo;
        *reinterpret_cast<float4*>(dgates + base + half + 4) = hi;
        bf16x8_k5 bv;
        bv[0] = (__bf16)lo.x; bv[1] = (__bf16)lo.y;
        bv[2] = (__bf16)lo.z; bv[3] = (__bf16)lo.w;
        bv[4] = (__bf16)hi.x; bv[5] = (__bf16)hi.y;
        bv[6] = (__bf16)hi.z; bv[7] = (__bf16)hi.w;
        *reinterpret_cast<bf16x8_k5*>(dgates_bf + base + half) = bv;
      }
    }
    __syncthreads();
    if (tid == 0) {
      __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __hip_atomic_fetch_add((gu32_t*)ctr, 1u, DRL_RLX_AGENT);
      const unsigned target = nblocks * (unsigned)(it + 1);
      unsigned spins = 0;
      while (__hip_atomic_load((gu32_t*)ctr, DRL_RLX_AGENT) < target) {
        __builtin_amdgcn_s_sleep(4);
        if (++spins > 5000000u) break;
      }
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    }
    __syncthreads();
  }
  // final dc (dL/dc0) back to global
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    int e = tid + u * 256;
    int row = e >> 4, hc = e & 15;
    if (row < B) dc0_out[(int64_t)row * H + n0 + hc] = dc_reg[u];
  }
}
}  // namespace

bool lstm_seq_fwd_bf16(torch::Tensor xp, torch::Tensor c0, torch::Tensor w_bf,
                       torch::Tensor hs, torch::Tensor cs, torch::Tensor h_bfs,
                       torch::Tensor acts, torch::Tensor tanhc,
                       torch::Tensor ctr) {
  int T = (int)xp.size(0), B = (int)xp.size(1), H = (int)c0.size(1);
  if (B > kLstmMaxB || H != 512) return false;
  constexpr int HH = 512;
  hipLaunchKernelGGL(lstm_seq_fwd_bf16_kernel<HH>, dim3(HH / 16), dim3(256), 0,
                     cur_stream(), xp.data_ptr<float>(), c0.data_ptr<float>(),
                     (const __bf16*)w_bf.data_ptr(), hs.data_ptr<float>(),
                     cs.data_ptr<float>(), (__bf16*)h_bfs.data_ptr(),
                     acts.data_ptr<float>(), tanhc.data_ptr<float>(),
                     (unsigned int*)ctr.data_ptr(), B, T);
  return true;
}

bool lstm_seq_bwd_bf16(torch::Tensor dh_init, torch::Tensor gout,
                       torch::Tensor dc_T, torch::Tensor w_t_bf,
                       torch::Tensor acts, torch::Tensor tanhc,
                       torch::Tensor cs, torch::Tensor dgates,
                       torch::Tensor dgates_bf, torch::Tensor dc0_out,
                       torch::Tensor ctr) {
  int T = (int)gout.size(0), B = (int)gout.size(1), H = (int)gout.size(2);
  if (B > kLstmMaxB || H != 512) return false;
  constexpr int HH = 512;
  hipLaunchKernelGGL(lstm_seq_bwd_bf16_kernel<HH>, dim3(HH / 16), dim3(256), 0,
                     cur_stream(), dh_init.data_ptr<float>(),
                     gout.data_ptr<float>(), dc_T.data_ptr<float>(),
                     (const __bf16*)w_t_bf.data_ptr(), acts.data_ptr<float>(),
                     tanhc.data_ptr<float>(), cs.data_ptr<float>(),
                     dgates.data_ptr<float>(), (__bf16*)dgates_bf.data_ptr(),
                     dc0_out.data_ptr<float>(),
                     (unsigned int*)ctr.data_ptr(), B, T);
  return true;
}

// ===========================================================================
// K3+K4+heads fused (round 2b): the dueling HEAD projections join the loss.
//   fwd: one wave per batch row computes adv_j = <h[:512], Wa_j> + ba_j and
//        val = <h[512:], Wv> + bv for all THREE forwards (online s, online
//        s', target s'), then dueling + n-step double-DQN TD inline.
//   bwd: dh closed form  dh[b,k<512] = -coef_b*gout*(Wa[a_b,k] - mean_j Wa[j,k])
//                        dh[b,512+k] = -coef_b*gout*Wv[k]
//        and a reduction kernel for (dWa, dba, dWv, dbv).
// Replaces the 6 head GEMM launches per step + the head-backward GEMM/
// reduce chain (~11 launches -> 3). h layout: the fused (B, 1024) hidden
// [adv-stream | val-stream] produced by the single w1 GEMM
// (ApexLearner._build_fast_forward).
// ===========================================================================
namespace {

// wave-level sum over 64 lanes
__device__ __forceinline__ float wave_sum(float v) {
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

template <int HH, int A>  // HH = per-stream hidden (512), A = actions (6)
__global__ __launch_bounds__(256) void dueling_q_loss_fwd_kernel(
    const __bf16* __restrict__ h_s,   // (B, 2*HH)
    const __bf16* __restrict__ h_on,  // (B, 2*HH)
    const __bf16* __restrict__ h_tg,  // (B, 2*HH)
    const __bf16* __restrict__ wa, const __bf16* __restrict__ ba,   // online
    const __bf16* __restrict__ wv, const __bf16* __restrict__ bv,
    const __bf16* __restrict__ wa_t, const __bf16* __restrict__ ba_t,  // target
    const __bf16* __restrict__ wv_t, const __bf16* __restrict__ bv_t,
    const int64_t* __restrict__ act, const float* __restrict__ rew,
    const float* __restrict__ done, const float* __restrict__ w, int B,
    float gamma_n, float alpha, float* __restrict__ loss_out,
    float* __restrict__ prio_out, float* __restrict__ grad_coef,
    float* __restrict__ qmax_out) {
  const int lane = threadIdx.x & 63;
  const int b = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (b >= B) return;
  const int k8 = lane * 8;  // this lane's 8-elem slice of each stream

  // per-lane fragments of the three hidden rows
  float ha_s[8], hv_s[8], ha_on[8], hv_on[8], ha_tg[8], hv_tg[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    ha_s[j] = (float)h_s[(int64_t)b * 2 * HH + k8 + j];
    hv_s[j] = (float)h_s[(int64_t)b * 2 * HH + HH + k8 + j];
    ha_on[j] = (float)h_on[(int64_t)b * 2 * HH + k8 + j];
    hv_on[j] = (float)h_on[(int64_t)b * 2 * HH + HH + k8 + j];
    ha_tg[j] = (float)h_tg[(int64_t)b * 2 * HH + k8 + j];
    hv_tg[j] = (float)h_tg[(int64_t)b * 2 * HH + HH + k8 + j];
  }
  float q_s[A], q_on[A], q_tg[A];  // adv dots (bias added on lane 0 later)
#pragma unroll
  for (int j = 0; j < A; ++j) {
    float pa_s = 0.f, pa_on = 0.f, pa_tg = 0.f;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const float wav = (float)wa[(int64_t)j * HH + k8 + u];
      const float wat = (float)wa_t[(int64_t)j * HH + k8 + u];
      pa_s += ha_s[u] * wav;
      pa_on += ha_on[u] * wav;
      pa_tg += ha_tg[u] * wat;
    }
    q_s[j] = wave_sum(pa_s);
    q_on[j] = wave_sum(pa_on);
    q_tg[j] = wave_sum(pa_tg);
  }
  float pv_s = 0.f, pv_on = 0.f, pv_tg = 0.f;
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    const float wvv = (float)wv[k8 + u];
    const float wvt = (float)wv_t[k8 + u];
    pv_s += hv_s[u] * wvv;
    pv_on += hv_on[u] * wvv;
    pv_tg += hv_tg[u] * wvt;
  }
  const float v_s = wave_sum(pv_s);
  const float v_tg = wave_sum(pv_tg);
  wave_sum(pv_on);  // unused (argmax needs adv only) — keep lanes converged

  if (lane == 0) {
    float mean_s = 0.f, mean_tg = 0.f, best = -1e30f;
    int a_star = 0;
#pragma unroll
    for (int j = 0; j < A; ++j) {
      const float adv_on = q_on[j] + (float)ba[j];
      if (adv_on > best) { best = adv_on; a_star = j; }
      q_s[j] += (float)ba[j];
      q_tg[j] += (float)ba_t[j];
      mean_s += q_s[j];
      mean_tg += q_tg[j];
    }
    mean_s /= A;
    mean_tg /= A;
    const float vs_full = v_s + (float)bv[0];
    const float vt_full = v_tg + (float)bv_t[0];
    const float q_tgt = q_tg[a_star] - mean_tg + vt_full;
    const float target = rew[b] + gamma_n * q_tgt * (1.0f - done[b]);
    float smax = q_s[0];
#pragma unroll
    for (int j = 1; j < A; ++j) smax = fmaxf(smax, q_s[j]);
    const float q = q_s[act[b]] - mean_s + vs_full;
    const float raw = target - q;
    const float td = fminf(1.0f, fmaxf(-1.0f, raw));
    prio_out[b] = __powf(fabsf(td) + 1e-7f, alpha);
    const float in_range = (raw > -1.0f && raw < 1.0f) ? 1.0f : 0.0f;
    const float invB = 1.0f / B;
    grad_coef[b] = w[b] * td * in_range * invB;
    atomicAdd(loss_out, 0.5f * w[b] * td * td * invB);
    atomicAdd(qmax_out, (smax - mean_s + vs_full) * invB);
  }
}

template <int HH, int A>
__global__ void dueling_q_loss_bwd_dh_kernel(
    const float* __restrict__ grad_coef, const int64_t* __restrict__ act,
    const float* __restrict__ gout, const __bf16* __restrict__ wa,
    const __bf16* __restrict__ wv, int B, __bf16* __restrict__ dh) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B * 2 * HH) return;
  const int b = i / (2 * HH);
  const int k = i - b * 2 * HH;
  const float c = -grad_coef[b] * gout[0];
  float v;
  if (k < HH) {
    float m = 0.f;
#pragma unroll
    for (int j = 0; j < A; ++j) m += (float)wa[(int64_t)j * HH + k];
    v = c * ((float)wa[(int64_t)act[b] * HH + k] - m / A);
  } else {
    v = c * (float)wv[k - HH];
  }
  dh[i] = (__bf16)v;
}

// phase 1: grid (A+1 head-rows, BS b-slices); block (j, s) accumulates its
// b-slice's partial dW row (+ bias partial) into the fp32 workspace
// ws[(A+1) rows x (HH+1)] x BS. phase 2 folds the BS partials and casts.
template <int HH, int A, int BS>
__global__ __launch_bounds__(256) void dueling_q_loss_bwd_dw_kernel(
    const float* __restrict__ grad_coef, const int64_t* __restrict__ act,
    const float* __restrict__ gout, const __bf16* __restrict__ h_s, int B,
    float* __restrict__ ws) {  // (BS, A+1, HH+1)
  const int j = blockIdx.x;
  const int sl = blockIdx.y;
  const int tid = threadIdx.x;
  const float g = gout[0];
  float accw[HH / 256];
#pragma unroll
  for (int u = 0; u < HH / 256; ++u) accw[u] = 0.f;
  float accb = 0.f;
  for (int b = sl; b < B; b += BS) {
    const float c = -grad_coef[b] * g;
    float alpha_jb;
    int64_t hoff;
    if (j < A) {
      alpha_jb = c * (((int)act[b] == j ? 1.0f : 0.0f) - 1.0f / A);
      hoff = (int64_t)b * 2 * HH;
    } else {
      alpha_jb = c;
      hoff = (int64_t)b * 2 * HH + HH;
    }
    accb += alpha_jb;
#pragma unroll
    for (int u = 0; u < HH / 256; ++u) {
      const int k = u * 256 + tid;
      accw[u] += alpha_jb * (float)h_s[hoff + k];
    }
  }
  float* row = ws + ((int64_t)sl * (A + 1) + j) * (HH + 1);
#pragma unroll
  for (int u = 0; u < HH / 256; ++u) row[u * 256 + tid] = accw[u];
  if (tid == 0) row[HH] = accb;
}

template <int HH, int A, int BS>
__global__ void dueling_q_loss_bwd_dw_fold_kernel(
    const float* __restrict__ ws, __bf16* __restrict__ dwa,
    __bf16* __restrict__ dba, __bf16* __restrict__ dwv,
    __bf16* __restrict__ dbv) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  constexpr int ROW = HH + 1;
  if (i >= (A + 1) * ROW) return;
  const int j = i / ROW;
  const int k = i - j * ROW;
  float v = 0.f;
#pragma unroll
  for (int s = 0; s < BS; ++s) v += ws[((int64_t)s * (A + 1) + j) * ROW + k];
  if (k < HH) {
    if (j < A)
      dwa[(int64_t)j * HH + k] = (__bf16)v;
    else
      dwv[k] = (__bf16)v;
  } else {
    if (j < A)
      dba[j] = (__bf16)v;
    else
      dbv[0] = (__bf16)v;
  }
}
}  // namespace

void dueling_q_loss_fwd(torch::Tensor h_s, torch::Tensor h_on,
                        torch::Tensor h_tg, torch::Tensor wa, torch::Tensor ba,
                        torch::Tensor wv, torch::Tensor bv, torch::Tensor wa_t,
                        torch::Tensor ba_t, torch::Tensor wv_t,
                        torch::Tensor bv_t, torch::Tensor act,
                        torch::Tensor rew, torch::Tensor done, torch::Tensor w,
                        double gamma_n, double alpha, torch::Tensor loss_out,
                        torch::Tensor prio_out, torch::Tensor grad_coef,
                        torch::Tensor qmax_out) {
  int B = (int)h_s.size(0);
  TORCH_CHECK(h_s.size(1) == 1024 && wa.size(0) == 6, "geometry: HH=512 A=6");
  hipLaunchKernelGGL((dueling_q_loss_fwd_kernel<512, 6>),
                     dim3((B + 3) / 4), dim3(256), 0, cur_stream(),
                     (const __bf16*)h_s.data_ptr(),
                     (const __bf16*)h_on.data_ptr(),
                     (const __bf16*)h_tg.data_ptr(),
                     (const __bf16*)wa.data_ptr(), (const __bf16*)ba.data_ptr(),
                     (const __bf16*)wv.data_ptr(), (const __bf16*)bv.data_ptr(),
                     (const __bf16*)wa_t.data_ptr(),
                     (const __bf16*)ba_t.data_ptr(),
                     (const __bf16*)wv_t.data_ptr(),
                     (const __bf16*)bv_t.data_ptr(), act.data_ptr<int64_t>(),
                     rew.data_ptr<float>(), done.data_ptr<float>(),
                     w.data_ptr<float>(), B, (float)gamma_n, (float)alpha,
                     loss_out.data_ptr<float>(), prio_out.data_ptr<float>(),
                     grad_coef.data_ptr<float>(), qmax_out.data_ptr<float>());
}

void dueling_q_loss_bwd(torch::Tensor grad_coef, torch::Tensor act,
                        torch::Tensor gout, torch::Tensor wa, torch::Tensor wv,
                        torch::Tensor h_s, torch::Tensor dh, torch::Tensor dwa,
                        torch::Tensor dba, torch::Tensor dwv,
                        torch::Tensor dbv) {
  int B = (int)h_s.size(0);
  hipLaunchKernelGGL((dueling_q_loss_bwd_dh_kernel<512, 6>),
                     dim3(ceil_div((int64_t)B * 1024, kBlock)), dim3(kBlock),
                     0, cur_stream(), grad_coef.data_ptr<float>(),
                     act.data_ptr<int64_t>(), gout.data_ptr<float>(),
                     (const __bf16*)wa.data_ptr(), (const __bf16*)wv.data_ptr(),
                     B, (__bf16*)dh.data_ptr());
  constexpr int BS = 32;  // b-slices (serial-B measured 16 us; 16 slices
                          // still left 32 serial iterations per block)
  auto ws = torch::empty({BS * 7 * 513}, torch::dtype(torch::kFloat32)
                                             .device(h_s.device()));
  hipLaunchKernelGGL((dueling_q_loss_bwd_dw_kernel<512, 6, BS>),
                     dim3(7, BS), dim3(256), 0, cur_stream(),
                     grad_coef.data_ptr<float>(), act.data_ptr<int64_t>(),
                     gout.data_ptr<float>(), (const __bf16*)h_s.data_ptr(), B,
                     ws.data_ptr<float>());
  hipLaunchKernelGGL((dueling_q_loss_bwd_dw_fold_kernel<512, 6, BS>),
                     dim3(ceil_div(7 * 513, kBlock)), dim3(kBlock), 0,
                     cur_stream(), ws.data_ptr<float>(),
                     (__bf16*)dwa.data_ptr(), (__bf16*)dba.data_ptr(),
                     (__bf16*)dwv.data_ptr(), (__bf16*)dbv.data_ptr());
}

// ===========================================================================
// K4/K6/K7 sequence fusion (round 2): R2D2's whole target/loss/priority
// pipeline in 3 kernels. The torch composition (nstep_recurrent_targets +
// IS-weighted loss + eta-mix priority) ran ~30 launches per step including
// an fp64 prefix-sum chain; here each (t, b) thread computes its truncated
// n-step return directly (n <= UNROLL_STEP <= 8 taps), double-DQN argmax
// over A inline, h/h^-1 value rescaling inline.
//   fwd:  td(W,B) + loss + value/|td| stats (wave-reduced atomics)
//   prio: eta-mix (0.9*max + 0.1*mean)^alpha over t per sequence
//   bwd:  dq_train(T-m, B, A) scatter in closed form
// Replaces R2D2/Learner.py:76-198 semantics (the action-slice
// defect is NOT replicated — window [MEM, T-1), truncated tail n).
// ===========================================================================
namespace {

__device__ __forceinline__ float vrescale(float x) {
  const float s = x >= 0.0f ? 1.0f : -1.0f;
  return s * (sqrtf(fabsf(x) + 1.0f) - 1.0f) + 1e-3f * x;
}

__device__ __forceinline__ float inv_vrescale(float x) {
  // closed-form inverse (R2D2/Learner.py:28-35); eps matches the repo-wide
  // _RESCALE_EPS = 1e-3 (ops/torch_ref.py:65)
  const float s = x >= 0.0f ? 1.0f : -1.0f;
  const float e = 1e-3f;
  const float t = 1.0f + 4.0f * e * (fabsf(x) + 1.0f + e);
  const float r = (sqrtf(t) - 1.0f) / (2.0f * e);
  return s * (r * r - 1.0f);
}

__global__ void r2d2_loss_fwd_kernel(
    const float* __restrict__ q_train,  // (T-m, B, A)
    const float* __restrict__ q_tgt,    // (T, B, A)
    const int* __restrict__ act,        // (T, B) int32
    const float* __restrict__ rew,      // (T, B)
    const float* __restrict__ done,     // (B)
    const float* __restrict__ w,        // (B)
    int T, int B, int A, int m, int n_step, float gamma, int rescale,
    float* __restrict__ td_out,         // (W, B), W = T-1-m
    float* __restrict__ stats)          // [loss, q_mean, td_abs] pre-zeroed
{
  const int W = T - 1 - m;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  float lc = 0.f, qc = 0.f, tc = 0.f;
  if (i < W * B) {
    const int ti = i / B;       // 0..W-1
    const int b = i - ti * B;
    const int t = m + ti;
    const int n = min(n_step, T - 1 - t);
    // discounted n-step reward sum
    float ret = 0.f, g = 1.0f;
    for (int k = 0; k < n; ++k) {
      ret += g * rew[(int64_t)(t + k) * B + b];
      g *= gamma;
    }
    const int tn = t + n;
    // double-DQN argmax at t+n over the ONLINE view: target rows pad the
    // burn-in region (never reached: tn >= m+1)
    const float* q_on_row = (tn >= m)
        ? q_train + ((int64_t)(tn - m) * B + b) * A
        : q_tgt + ((int64_t)tn * B + b) * A;
    int a_star = 0;
    float best = q_on_row[0];
    for (int a = 1; a < A; ++a)
      if (q_on_row[a] > best) { best = q_on_row[a]; a_star = a; }
    float q_boot = q_tgt[((int64_t)tn * B + b) * A + a_star];
    if (rescale) q_boot = inv_vrescale(q_boot);
    if (tn == T - 1) q_boot *= (1.0f - done[b]);
    float G = ret + g * q_boot;  // g == gamma^n
    if (rescale) G = vrescale(G);
    const float q_taken =
        q_train[((int64_t)ti * B + b) * A + act[(int64_t)t * B + b]];
    const float td = G - q_taken;
    td_out[(int64_t)ti * B + b] = td;
    const float inv = 1.0f / (B * W);
    lc = 0.5f * w[b] * td * td * inv;
    qc = q_taken * inv;
    tc = fabsf(td) * inv;
  }
  for (int off = 32; off > 0; off >>= 1) {
    lc += __shfl_down(lc, off, 64);
    qc += __shfl_down(qc, off, 64);
    tc += __shfl_down(tc, off, 64);
  }
  if ((threadIdx.x & 63) == 0 && (lc != 0.f || qc != 0.f || tc != 0.f)) {
    atomicAdd(&stats[0], lc);
    atomicAdd(&stats[1], qc);
    atomicAdd(&stats[2], tc);
  }
}

// one wave per sequence b: prio_b = (eta*max_t|td| + (1-eta)*mean_t|td|)^alpha
__global__ void r2d2_prio_kernel(const float* __restrict__ td, int W, int B,
                                 float alpha, float eta,
                                 float* __restrict__ prio) {
  const int b = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (b >= B) return;
  const int lane = threadIdx.x & 63;
  float mx = 0.f, sm = 0.f;
  for (int t = lane; t < W; t += 64) {
    const float v = fabsf(td[(int64_t)t * B + b]);
    mx = fmaxf(mx, v);
    sm += v;
  }
  for (int off = 32; off > 0; off >>= 1) {
    mx = fmaxf(mx, __shfl_down(mx, off, 64));
    sm += __shfl_down(sm, off, 64);
  }
  if (lane == 0)
    prio[b] = __powf(eta * mx + (1.0f - eta) * sm / W, alpha);
}

__global__ void r2d2_loss_bwd_kernel(const float* __restrict__ td,
                                     const int* __restrict__ act,
                                     const float* __restrict__ w,
                                     const float* __restrict__ gout,
                                     int T, int B, int A, int m,
                                     float* __restrict__ dq)  // (T-m, B, A)
{
  const int W = T - 1 - m;
  const int64_t total = (int64_t)(T - m) * B * A;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  const int a = (int)(i % A);
  const int64_t t2 = i / A;
  const int b = (int)(t2 % B);
  const int ti = (int)(t2 / B);
  float v = 0.f;
  if (ti < W && a == act[(int64_t)(m + ti) * B + b])
    v = -w[b] * td[(int64_t)ti * B + b] * gout[0] / (B * W);
  dq[i] = v;
}
}  // namespace

void r2d2_loss_fwd(torch::Tensor q_train, torch::Tensor q_tgt,
                   torch::Tensor act, torch::Tensor rew, torch::Tensor done,
                   torch::Tensor w, int64_t m, int64_t n_step, double gamma,
                   bool rescale, torch::Tensor td_out, torch::Tensor stats) {
  int T = (int)q_tgt.size(0), B = (int)q_tgt.size(1), A = (int)q_tgt.size(2);
  const int W = T - 1 - (int)m;
  hipLaunchKernelGGL(r2d2_loss_fwd_kernel,
                     dim3(ceil_div((int64_t)W * B, kBlock)), dim3(kBlock), 0,
                     cur_stream(), q_train.data_ptr<float>(),
                     q_tgt.data_ptr<float>(), act.data_ptr<int>(),
                     rew.data_ptr<float>(), done.data_ptr<float>(),
                     w.data_ptr<float>(), T, B, A, (int)m, (int)n_step,
                     (float)gamma, rescale ? 1 : 0, td_out.data_ptr<float>(),
                     stats.data_ptr<float>());
}

void r2d2_prio(torch::Tensor td, double alpha, double eta,
               torch::Tensor prio) {
  int W = (int)td.size(0), B = (int)td.size(1);
  hipLaunchKernelGGL(r2d2_prio_kernel, dim3(ceil_div(B, 4)), dim3(256), 0,
                     cur_stream(), td.data_ptr<float>(), W, B, (float)alpha,
                     (float)eta, prio.data_ptr<float>());
}

void r2d2_loss_bwd(torch::Tensor td, torch::Tensor act, torch::Tensor w,
                   torch::Tensor gout, int64_t T, int64_t m,
                   torch::Tensor dq) {
  int B = (int)td.size(1), A = (int)dq.size(2);
  const int64_t total = (int64_t)(T - m) * B * A;
  hipLaunchKernelGGL(r2d2_loss_bwd_kernel, dim3(ceil_div(total, kBlock)),
                     dim3(kBlock), 0, cur_stream(), td.data_ptr<float>(),
                     act.data_ptr<int>(), w.data_ptr<float>(),
                     gout.data_ptr<float>(), (int)T, B, A, (int)m,
                     dq.data_ptr<float>());
}

// ===========================================================================
// Sequence transpose (R2D2): (B, T, R bytes) -> (T, B, R) row gather. The
// torch permute().contiguous() of the 72 MB uint8 frame block ran at
// ~1.5 TB/s (92.7 us); whole-row uint4 block copies stream at full rate.
// ===========================================================================
namespace {
__global__ void seq_transpose_rows_kernel(const uint4* __restrict__ src,
                                          uint4* __restrict__ dst, int B,
                                          int T, int r16) {
  const int bt = blockIdx.x;
  const int b = bt / T;
  const int t = bt - b * T;
  const uint4* s = src + (int64_t)bt * r16;
  uint4* d = dst + ((int64_t)t * B + b) * r16;
  for (int i = threadIdx.x; i < r16; i += blockDim.x) d[i] = s[i];
}
}  // namespace

void seq_transpose_rows(torch::Tensor src, torch::Tensor dst) {
  // src: (B, T, ...) contiguous; dst: (T, B, ...) contiguous, same dtype
  const int B = (int)src.size(0), T = (int)src.size(1);
  const int64_t row_bytes =
      src.numel() * src.element_size() / ((int64_t)B * T);
  TORCH_CHECK(row_bytes % 16 == 0, "row bytes must be 16-aligned");
  TORCH_CHECK(src.is_contiguous() && dst.is_contiguous());
  hipLaunchKernelGGL(seq_transpose_rows_kernel, dim3(B * T), dim3(256), 0,
                     cur_stream(), (const uint4*)src.data_ptr(),
                     (uint4*)dst.data_ptr(), B, T, (int)(row_bytes / 16));
}

// ===========================================================================
// Fused replay gather (K10): one launch copies every column's sampled rows
// (the generic per-column index_select chain was 8-10 kernels per sample:
// 2 big frame gathers + small-column gathers/scatters, ~50 us/step).
// Row bytes that are 16-aligned stream as uint4; ragged columns fall back
// to byte copies. Block (b, c) copies row idx[b] of column c.
// ===========================================================================
namespace {
struct GatherCols {
  const uint8_t* src[8];
  uint8_t* dst[8];
  int64_t row_bytes[8];
  int ncols;
};

__global__ void gather_rows_kernel(GatherCols a,
                                   const int64_t* __restrict__ idx, int k,
                                   int chunks) {
  const int b = blockIdx.x;
  const int c = blockIdx.y;
  const int ch = blockIdx.z;  // big rows split across chunk blocks (a
                              // 564 KB R2D2 row on ONE block starved the
                              // chip: 32 blocks for an 18 MB gather)
  if (b >= k || c >= a.ncols) return;
  const int64_t rb = a.row_bytes[c];
  const uint8_t* s = a.src[c] + idx[b] * rb;
  uint8_t* d = a.dst[c] + (int64_t)b * rb;
  if ((rb & 15) == 0) {
    const int64_t n16 = rb / 16;
    const int64_t per = (n16 + chunks - 1) / chunks;
    const int64_t lo = ch * per;
    const int64_t hi = lo + per < n16 ? lo + per : n16;
    const uint4* s4 = reinterpret_cast<const uint4*>(s);
    uint4* d4 = reinterpret_cast<uint4*>(d);
    for (int64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) d4[i] = s4[i];
  } else if (ch == 0) {
    for (int64_t i = threadIdx.x; i < rb; i += blockDim.x) d[i] = s[i];
  }
}
}  // namespace

void gather_rows(torch::Tensor idx, std::vector<torch::Tensor> srcs,
                 std::vector<torch::Tensor> dsts) {
  TORCH_CHECK(srcs.size() == dsts.size() && srcs.size() <= 8);
  GatherCols a{};
  a.ncols = (int)srcs.size();
  for (int c = 0; c < a.ncols; ++c) {
    TORCH_CHECK(srcs[c].is_contiguous() && dsts[c].is_contiguous());
    a.src[c] = (const uint8_t*)srcs[c].data_ptr();
    a.dst[c] = (uint8_t*)dsts[c].data_ptr();
    a.row_bytes[c] =
        srcs[c].numel() * srcs[c].element_size() / srcs[c].size(0);
  }
  const int k = (int)idx.numel();
  int64_t max_rb = 0;
  for (int c = 0; c < a.ncols; ++c)
    max_rb = a.row_bytes[c] > max_rb ? a.row_bytes[c] : max_rb;
  // target >= ~1024 blocks for the biggest column
  int chunks = 1;
  while ((int64_t)k * chunks < 1024 && (max_rb / 16) / chunks > 256)
    chunks *= 2;
  hipLaunchKernelGGL(gather_rows_kernel, dim3(k, a.ncols, chunks), dim3(256),
                     0, cur_stream(), a, idx.data_ptr<int64_t>(), k, chunks);
}

// ===========================================================================
// Diagnostic: persistent-kernel step-cost decomposition (tools/
// gpu_dgrad_bench.py). mode 0 = grid barrier only; 1 = + h burst-stage;
// 2 = + MFMA phase over LDS; isolates what the ~8 us/step persistent LSTM
// floor is made of.
// ===========================================================================
namespace {
template <int H>
__global__ __launch_bounds__(256) void lstm_diag_kernel(
    const __bf16* __restrict__ h_bfs, const __bf16* __restrict__ w_bf,
    float* __restrict__ sink, unsigned int* __restrict__ ctr, int B, int T,
    int mode) {
  constexpr int LROW = H + 32;
  __shared__ __bf16 wlds[64][LROW];
  __shared__ __bf16 hlds[32][LROW];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int s16 = blockIdx.x * 16;
  const int kbase = (lane >> 4) * 8;
  const unsigned nblocks = gridDim.x;
  if (mode >= 2) {
    for (int base = tid * 8; base < 64 * H; base += 256 * 8) {
      const int r = base / H;
      const int g = r >> 4, c = r & 15;
      *reinterpret_cast<bf16x8_k5*>(&wlds[r][base - r * H]) =
          *reinterpret_cast<const bf16x8_k5*>(
              w_bf + ((int64_t)g * H + s16 + c) * H + (base - r * H));
    }
  }
  __syncthreads();
  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = {0.f, 0.f, 0.f, 0.f};
  for (int t = 0; t < T; ++t) {
    if (mode >= 1) {
      const __bf16* h_bf = h_bfs + (int64_t)(t & 1) * B * H;
      for (int base = tid * 8; base < 32 * H; base += 256 * 8) {
        const int r = base / H;
        bf16x8_k5 v{};
        if (r < B)
          v = *reinterpret_cast<const bf16x8_k5*>(h_bf + (int64_t)r * H +
                                                  (base - r * H));
        *reinterpret_cast<bf16x8_k5*>(&hlds[r][base - r * H]) = v;
      }
      __syncthreads();
    }
    if (mode >= 2) {
#pragma unroll 4
      for (int kb = 0; kb < H; kb += 32) {
        bf16x8_k5 bfrag = *reinterpret_cast<const bf16x8_k5*>(
            &wlds[wave * 16 + (lane & 15)][kb + kbase]);
        bf16x8_k5 a0 = *reinterpret_cast<const bf16x8_k5*>(
            &hlds[lane & 15][kb + kbase]);
        bf16x8_k5 a1 = *reinterpret_cast<const bf16x8_k5*>(
            &hlds[16 + (lane & 15)][kb + kbase]);
        acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bfrag, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, bfrag, acc1, 0, 0, 0);
      }
    }
    __syncthreads();
    if (tid == 0) {
      __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __hip_atomic_fetch_add((gu32_t*)ctr, 1u, DRL_RLX_AGENT);
      const unsigned target = nblocks * (unsigned)(t + 1);
      unsigned spins = 0;
      while (__hip_atomic_load((gu32_t*)ctr, DRL_RLX_AGENT) < target) {
        __builtin_amdgcn_s_sleep(4);
        if (++spins > 5000000u) break;
      }
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    }
    __syncthreads();
  }
  if (tid == 0 && acc0[0] > 1e30f) sink[blockIdx.x] = acc0[0] + acc1[0];
}
}  // namespace

void lstm_diag(torch::Tensor h_bfs, torch::Tensor w_bf, torch::Tensor sink,
               torch::Tensor ctr, int64_t B, int64_t T, int64_t mode) {
  hipLaunchKernelGGL(lstm_diag_kernel<512>, dim3(32), dim3(256), 0,
                     cur_stream(), (const __bf16*)h_bfs.data_ptr(),
                     (const __bf16*)w_bf.data_ptr(), sink.data_ptr<float>(),
                     (unsigned int*)ctr.data_ptr(), (int)B, (int)T,
                     (int)mode);
}
