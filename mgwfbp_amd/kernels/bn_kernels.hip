// NHWC BatchNorm (training fwd/bwd + inference) for gfx950 / CDNA4.
//
// Why: rocprof on the ResNet-50 bf16 step (profiles/
// resnet50_n1_bf16_kernel_stats.md) shows MIOpen's spatial BatchNorm --
// 6 kernels per layer, run in fp32 with bf16<->fp32 cast kernels around
// them (autocast promotes BN to fp32) -- costing 34% of the step. This
// implementation consumes bf16/fp16 NHWC activations DIRECTLY with fp32
// accumulation: ~2.5x fewer HBM bytes and no cast kernels.
//
// Layout contract: x is channels_last (N,H,W,C) dense, viewed as
// rows = N*H*W by C. C must be a multiple of 8.
//
// Thread mapping (all kernels): a block of 256 threads is a
// (rows_per_block x octets_per_row) tile where octets_per_row = C/8
// capped at 256: each active thread owns EIGHT consecutive channels
// (one 16-byte bf16 octet -> coalesced 16 B/lane loads, guide G13) and
// loops over rows with per-channel params held in registers.
//
// Reduction is TWO-STAGE with a per-block partials buffer -- register
// accumulators -> LDS across the block's rows -> ONE plain store of the
// block's 2C partial sums -> a finalize kernel sums over blocks. No
// global atomics anywhere: a first version used one atomicAdd per
// channel per block and the same-address serialization made small
// layers ~30x slower than the data movement (guide G12's "reduce then
// one atomic" still serializes when blocks >> channels).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define CHECK_HIP(x)                                                         \
  do {                                                                       \
    hipError_t err__ = (x);                                                  \
    TORCH_CHECK(err__ == hipSuccess, "HIP error: ",                          \
                hipGetErrorString(err__));                                   \
  } while (0)

namespace {

constexpr int kBlock = 256;

template <typename T> struct VecIO;
template <> struct VecIO<float> {
  static __device__ inline void load(const float* p, float v[8]) {
    const float4 a = *reinterpret_cast<const float4*>(p);
    const float4 b = *reinterpret_cast<const float4*>(p + 4);
    v[0] = a.x; v[1] = a.y; v[2] = a.z; v[3] = a.w;
    v[4] = b.x; v[5] = b.y; v[6] = b.z; v[7] = b.w;
  }
  static __device__ inline void store(float* p, const float v[8]) {
    *reinterpret_cast<float4*>(p) = {v[0], v[1], v[2], v[3]};
    *reinterpret_cast<float4*>(p + 4) = {v[4], v[5], v[6], v[7]};
  }
};
template <> struct VecIO<__hip_bfloat16> {
  static __device__ inline void load(const __hip_bfloat16* p, float v[8]) {
    const uint4 raw = *reinterpret_cast<const uint4*>(p);
    const unsigned u[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const __hip_bfloat162 h2 =
          *reinterpret_cast<const __hip_bfloat162*>(&u[i]);
      v[2 * i] = __bfloat162float(h2.x);
      v[2 * i + 1] = __bfloat162float(h2.y);
    }
  }
  static __device__ inline void store(__hip_bfloat16* p, const float v[8]) {
    uint4 raw;
    unsigned* u = reinterpret_cast<unsigned*>(&raw);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __hip_bfloat162 h2 = {__float2bfloat16(v[2 * i]),
                            __float2bfloat16(v[2 * i + 1])};
      u[i] = *reinterpret_cast<unsigned*>(&h2);
    }
    *reinterpret_cast<uint4*>(p) = raw;
  }
};
template <> struct VecIO<__half> {
  static __device__ inline void load(const __half* p, float v[8]) {
    const uint4 raw = *reinterpret_cast<const uint4*>(p);
    const unsigned u[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const __half2 h2 = *reinterpret_cast<const __half2*>(&u[i]);
      v[2 * i] = __half2float(h2.x);
      v[2 * i + 1] = __half2float(h2.y);
    }
  }
  static __device__ inline void store(__half* p, const float v[8]) {
    uint4 raw;
    unsigned* u = reinterpret_cast<unsigned*>(&raw);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __half2 h2 = __floats2half2_rn(v[2 * i], v[2 * i + 1]);
      u[i] = *reinterpret_cast<unsigned*>(&h2);
    }
    *reinterpret_cast<uint4*>(p) = raw;
  }
};

__device__ inline void tile_map(long C, int& opr, int& rpb, int& o,
                                int& rl, bool& active) {
  opr = (int)min(C / 8, (long)kBlock);
  rpb = kBlock / opr;
  o = threadIdx.x % opr;
  rl = threadIdx.x / opr;
  active = (rl < rpb);
}

// Generic per-block reduction of two 8-wide accumulators (a, b) keyed by
// channel octet, then ONE plain store per channel into
// partial[blockIdx.x * 2C .. +2C) (sum_a in [0,C), sum_b in [C,2C)).
__device__ inline void block_reduce_store(float a[8], float b[8], long C,
                                          int opr, int rpb, int o, int rl,
                                          bool active, long c0_wide,
                                          bool wide,
                                          float* __restrict__ partial) {
  float* out = partial + (long)blockIdx.x * 2 * C;
  if (wide) {
    // C > 2048: each octet owned by exactly one thread in the block
    if (active) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        out[c0_wide + i] = a[i];
        out[C + c0_wide + i] = b[i];
      }
    }
    return;
  }
  __shared__ float lds[kBlock * 16];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    lds[(rl * opr + o) * 16 + i] = active ? a[i] : 0.f;
    lds[(rl * opr + o) * 16 + 8 + i] = active ? b[i] : 0.f;
  }
  __syncthreads();
  if (rl == 0 && active) {
    float fa[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    float fb[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < rpb; ++r) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        fa[i] += lds[(r * opr + o) * 16 + i];
        fb[i] += lds[(r * opr + o) * 16 + 8 + i];
      }
    }
    const long c0 = (long)o * 8;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      out[c0 + i] = fa[i];
      out[C + c0 + i] = fb[i];
    }
  }
}

// -------------------------------------------------- fwd: reduce ---------
template <typename T>
__global__ __launch_bounds__(kBlock) void bn_fwd_reduce_kernel(
    const T* __restrict__ x, long rows, long C,
    float* __restrict__ partial) {
  int opr, rpb, o, rl; bool active;
  tile_map(C, opr, rpb, o, rl, active);
  const long octets = C / 8;
  const bool wide = octets > opr;
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float q[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  long c0_last = 0;
  if (active) {
    const long rstride = (long)gridDim.x * rpb;
    for (long ob = o; ob < octets; ob += opr) {
      const long c0 = ob * 8;
      c0_last = c0;
      long r = (long)blockIdx.x * rpb + rl;
      // 4x unrolled: four independent 16B loads in flight per thread
      // (the rolled loop compiles to load -> s_waitcnt vmcnt(0) -> use,
      // one load outstanding = latency-bound)
      for (; r + 3 * rstride < rows; r += 4 * rstride) {
        float v0[8], v1[8], v2[8], v3[8];
        VecIO<T>::load(x + r * C + c0, v0);
        VecIO<T>::load(x + (r + rstride) * C + c0, v1);
        VecIO<T>::load(x + (r + 2 * rstride) * C + c0, v2);
        VecIO<T>::load(x + (r + 3 * rstride) * C + c0, v3);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          s[i] += v0[i] + v1[i] + v2[i] + v3[i];
          q[i] += v0[i] * v0[i] + v1[i] * v1[i] + v2[i] * v2[i]
                  + v3[i] * v3[i];
        }
      }
      for (; r < rows; r += rstride) {
        float v[8];
        VecIO<T>::load(x + r * C + c0, v);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          s[i] += v[i];
          q[i] += v[i] * v[i];
        }
      }
      if (wide) {
        float* out = partial + (long)blockIdx.x * 2 * C;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          out[c0 + i] = s[i];
          out[C + c0 + i] = q[i];
          s[i] = 0; q[i] = 0;
        }
      }
    }
  }
  if (!wide)
    block_reduce_store(s, q, C, opr, rpb, o, rl, active, c0_last, false,
                       partial);
}

// ------------------------------------------------ finalize --------------
// Sum the [nblk][2C] partials. A first version used ceil(C/256) blocks
// each looping all nblk rows: 1-8 blocks of latency-bound strided loads
// = 200+ us, 3x the cost of the actual data passes. Now each block owns
// kFinC channels and its 256 threads split the nblk rows 8 ways, with
// an LDS combine — parallelism 8C instead of C.
constexpr int kFinC = 32;                  // channels per finalize block
constexpr int kFinG = kBlock / kFinC;      // row-subsets per block

__device__ inline void finalize_sums(const float* __restrict__ partial,
                                     long b0, long b1, long C, long& c,
                                     bool& leader, float& s, float& q) {
  const int cl = threadIdx.x % kFinC;
  const int g = threadIdx.x / kFinC;
  c = (long)blockIdx.x * kFinC + cl;
  s = 0.f; q = 0.f;
  if (c < C) {
    for (long b = b0 + g; b < b1; b += kFinG) {
      s += partial[b * 2 * C + c];
      q += partial[b * 2 * C + C + c];
    }
  }
  __shared__ float lds[kBlock * 2];
  lds[threadIdx.x] = s;
  lds[kBlock + threadIdx.x] = q;
  __syncthreads();
  leader = (g == 0 && c < C);
  if (leader) {
    for (int gg = 1; gg < kFinG; ++gg) {
      s += lds[gg * kFinC + cl];
      q += lds[kBlock + gg * kFinC + cl];
    }
  }
}

// Two-level tree for large nblk: stage1 splits the block range
// gridDim.y ways (PMC: a single-level finalize at nblk=1024, C=64 is
// 33 us latency-bound — 128 strided loads per thread on 2 blocks).
__global__ __launch_bounds__(kBlock) void bn_finalize_stage1_kernel(
    const float* __restrict__ partial, long nblk, long C,
    float* __restrict__ stage) {
  const long chunk = (nblk + gridDim.y - 1) / gridDim.y;
  const long b0 = (long)blockIdx.y * chunk;
  const long b1 = b0 + chunk < nblk ? b0 + chunk : nblk;
  long c; bool leader; float s, q;
  finalize_sums(partial, b0, b1, C, c, leader, s, q);
  if (!leader) return;
  stage[(long)blockIdx.y * 2 * C + c] = s;
  stage[(long)blockIdx.y * 2 * C + C + c] = q;
}

__global__ __launch_bounds__(kBlock) void bn_fwd_finalize_kernel(
    const float* __restrict__ partial, long nblk, long C, float M,
    float eps, float momentum, float* __restrict__ mean,
    float* __restrict__ invstd, float* __restrict__ running_mean,
    float* __restrict__ running_var) {
  long c; bool leader; float s, q;
  finalize_sums(partial, 0, nblk, C, c, leader, s, q);
  if (!leader) return;
  const float m = s / M;
  float var = q / M - m * m;
  var = var < 0.f ? 0.f : var;
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    const float unbiased = M > 1.f ? var * M / (M - 1.f) : var;
    running_var[c] =
        (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// ------------------------------------- fused finalize (small layers) ----
// For small inputs (rblocks <= kFusedFinBlocks, C <= 2048 so one block's
// octet tiling covers all channels) the separate finalize kernel is pure
// launch floor (~5 us for reading a few KB — resnet20's graph-replayed
// step is ~560 such launches). Each norm/dx block re-derives the channel
// sums from the partials directly in its per-octet preamble (<= 64x16
// L2-hot loads per octet) and block 0 writes the stat tensors; the
// finalize launch disappears. 6 -> 4 kernels per BN layer fwd+bwd.

constexpr int kFusedFinBlocks = 64;

// Runtime switch, DEFAULT OFF: same-box A/B (resnet20 bs32, 300-step
// graph replay, 2026-09-14) measured the fused-finalize variant 9%
// SLOWER end-to-end (1.547 vs 1.412 ms/step) both with a serial leader
// preamble and with the cooperative LDS-tree one — the 32 KB LDS
// footprint + preamble barrier cost the streaming loop more than the
// ~4.6 us finalize launch it saves. Kept behind MGX_BN_FUSED_FIN=1 for
// re-evaluation on future ROCm versions.
inline bool fused_fin_enabled() {
  static const bool on = [] {
    const char* v = getenv("MGX_BN_FUSED_FIN");
    return v != nullptr && v[0] == '1';
  }();
  return on;
}

// Block-cooperative finalize of the per-octet channel sums: the rpb
// row-threads of each octet column split the nblk partial rows, then an
// LDS tree reduces to the rl==0 leader (a serial leader loop measured
// 9% SLOWER end-to-end on resnet20 — every block stalled ~10us in the
// preamble). Returns true on the leader lane with (s, q) final.
__device__ inline bool partial_sums_coop(
    const float* __restrict__ partial, long nblk, long C, long c0,
    int opr, int rpb, int o, int rl, bool active,
    float* __restrict__ lds, float s[8], float q[8]) {
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    s[i] = 0.f;
    q[i] = 0.f;
  }
  if (active) {
    for (long b = rl; b < nblk; b += rpb) {
      const float* row = partial + b * 2 * C;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        s[i] += row[c0 + i];
        q[i] += row[C + c0 + i];
      }
    }
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    lds[(rl * opr + o) * 16 + i] = active ? s[i] : 0.f;
    lds[(rl * opr + o) * 16 + 8 + i] = active ? q[i] : 0.f;
  }
  __syncthreads();
  if (rl != 0 || !active) return false;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    s[i] = 0.f;
    q[i] = 0.f;
  }
  for (int r = 0; r < rpb; ++r) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      s[i] += lds[(r * opr + o) * 16 + i];
      q[i] += lds[(r * opr + o) * 16 + 8 + i];
    }
  }
  return true;
}

// Host guarantees opr == octets here (C <= 2048), so every thread takes
// exactly ONE octet and the LDS barrier is uniform. One leader thread
// per octet (rl == 0) derives the stats from the partials; the rest
// read them from LDS — without this the preamble is redundantly
// recomputed rpb times per block (measured 7% REGRESSION on resnet20
// before the leader/LDS split).
template <typename T, bool ADD>
__global__ __launch_bounds__(kBlock) void bn_fwd_norm_fin_kernel(
    const T* __restrict__ x, const T* __restrict__ res, T* __restrict__ y,
    long rows, long C, const float* __restrict__ partial, long nblk,
    float M, float eps, float momentum, const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ mean_out,
    float* __restrict__ invstd_out, float* __restrict__ running_mean,
    float* __restrict__ running_var, bool relu) {
  int opr, rpb, o, rl; bool active;
  tile_map(C, opr, rpb, o, rl, active);
  const long c0 = (long)o * 8;
  __shared__ float lds[kBlock * 16];
  __shared__ float smu[kBlock * 8], sis[kBlock * 8];   // [octet*8+i]
  {
    float s[8], q[8];
    const bool leader = partial_sums_coop(partial, nblk, C, c0, opr, rpb,
                                          o, rl, active, lds, s, q);
    if (leader) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float m = s[i] / M;
        float var = q[i] / M - m * m;
        var = var < 0.f ? 0.f : var;
        const float is = rsqrtf(var + eps);
        smu[o * 8 + i] = m;
        sis[o * 8 + i] = is;
        if (blockIdx.x == 0) {
          mean_out[c0 + i] = m;
          invstd_out[c0 + i] = is;
          if (running_mean != nullptr) {
            running_mean[c0 + i] =
                (1.f - momentum) * running_mean[c0 + i] + momentum * m;
            const float unbiased = M > 1.f ? var * M / (M - 1.f) : var;
            running_var[c0 + i] = (1.f - momentum) * running_var[c0 + i]
                                  + momentum * unbiased;
          }
        }
      }
    }
  }
  __syncthreads();
  if (!active) return;
  float sc[8], sh[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const float g = gamma ? gamma[c0 + i] : 1.f;
    sc[i] = g * sis[o * 8 + i];
    sh[i] = (beta ? beta[c0 + i] : 0.f) - smu[o * 8 + i] * sc[i];
  }
  const long rstride = (long)gridDim.x * rpb;
  long r = (long)blockIdx.x * rpb + rl;
  for (; r + 3 * rstride < rows; r += 4 * rstride) {
    float v[4][8], a[4][8];
#pragma unroll
    for (int u = 0; u < 4; ++u)
      VecIO<T>::load(x + (r + u * rstride) * C + c0, v[u]);
    if (ADD) {
#pragma unroll
      for (int u = 0; u < 4; ++u)
        VecIO<T>::load(res + (r + u * rstride) * C + c0, a[u]);
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float t = v[u][i] * sc[i] + sh[i];
        if (ADD) t += a[u][i];
        if (relu) t = t > 0.f ? t : 0.f;
        v[u][i] = t;
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u)
      VecIO<T>::store(y + (r + u * rstride) * C + c0, v[u]);
  }
  for (; r < rows; r += rstride) {
    float v[8], a[8];
    VecIO<T>::load(x + r * C + c0, v);
    if (ADD) VecIO<T>::load(res + r * C + c0, a);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      v[i] = v[i] * sc[i] + sh[i];
      if (ADD) v[i] += a[i];
      if (relu) v[i] = v[i] > 0.f ? v[i] : 0.f;
    }
    VecIO<T>::store(y + r * C + c0, v);
  }
}

template <typename T, bool RELU, bool ADD>
__global__ __launch_bounds__(kBlock) void bn_bwd_dx_fin_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ res, T* __restrict__ dx, T* __restrict__ dres,
    long rows, long C, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, const float* __restrict__ partial,
    long nblk, float invM, float* __restrict__ dbeta_out,
    float* __restrict__ dgamma_out) {
  int opr, rpb, o, rl; bool active;
  tile_map(C, opr, rpb, o, rl, active);
  const long c0 = (long)o * 8;
  __shared__ float lds[kBlock * 16];
  __shared__ float ssd[kBlock * 8], ssx[kBlock * 8];   // [octet*8+i]
  {
    float sd[8], sx[8];
    const bool leader = partial_sums_coop(partial, nblk, C, c0, opr, rpb,
                                          o, rl, active, lds, sd, sx);
    if (leader) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        ssd[o * 8 + i] = sd[i];
        ssx[o * 8 + i] = sx[i];
        if (blockIdx.x == 0) {
          dbeta_out[c0 + i] = sd[i];
          dgamma_out[c0 + i] = sx[i];
        }
      }
    }
  }
  __syncthreads();
  if (!active) return;
  float mu[8], is[8], gi[8], md[8], mx[8], ga[8], be[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    mu[i] = mean[c0 + i];
    is[i] = invstd[c0 + i];
    ga[i] = gamma ? gamma[c0 + i] : 1.f;
    be[i] = beta ? beta[c0 + i] : 0.f;
    gi[i] = ga[i] * is[i];
    md[i] = ssd[o * 8 + i] * invM;
    mx[i] = ssx[o * 8 + i] * invM;
  }
  const long rstride = (long)gridDim.x * rpb;
  long r = (long)blockIdx.x * rpb + rl;
  constexpr int U = ADD ? 2 : 4;
  for (; r + (U - 1) * rstride < rows; r += U * rstride) {
    float g[U][8], v[U][8], a[U][8];
#pragma unroll
    for (int u = 0; u < U; ++u) {
      VecIO<T>::load(dy + (r + u * rstride) * C + c0, g[u]);
      VecIO<T>::load(x + (r + u * rstride) * C + c0, v[u]);
      if (ADD) VecIO<T>::load(res + (r + u * rstride) * C + c0, a[u]);
    }
#pragma unroll
    for (int u = 0; u < U; ++u) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float xh = (v[u][i] - mu[i]) * is[i];
        if (RELU) {
          const float rr = ADD ? a[u][i] : 0.f;
          if (xh * ga[i] + be[i] + rr <= 0.f) g[u][i] = 0.f;
        }
        if (ADD) a[u][i] = g[u][i];
        g[u][i] = gi[i] * (g[u][i] - md[i] - xh * mx[i]);
      }
    }
#pragma unroll
    for (int u = 0; u < U; ++u) {
      if (ADD) VecIO<T>::store(dres + (r + u * rstride) * C + c0, a[u]);
      VecIO<T>::store(dx + (r + u * rstride) * C + c0, g[u]);
    }
  }
  for (; r < rows; r += rstride) {
    float g[8], v[8], a[8];
    VecIO<T>::load(dy + r * C + c0, g);
    VecIO<T>::load(x + r * C + c0, v);
    if (ADD) VecIO<T>::load(res + r * C + c0, a);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float xh = (v[i] - mu[i]) * is[i];
      if (RELU && xh * ga[i] + be[i] + (ADD ? a[i] : 0.f) <= 0.f)
        g[i] = 0.f;
      if (ADD) a[i] = g[i];
      g[i] = gi[i] * (g[i] - md[i] - xh * mx[i]);
    }
    if (ADD) VecIO<T>::store(dres + r * C + c0, a);
    VecIO<T>::store(dx + r * C + c0, g);
  }
}

// --------------------------------------------------- fwd: norm ----------
// ADD: y = [relu](bn(x) + res) — the ResNet post-add activation folded
// into the normalize pass: ONE kernel replaces bn_out store + add read/
// read/write + relu read/write (4 extra full-tensor HBM passes in the
// unfused eager chain). res is loaded once per element alongside x.
template <typename T, bool ADD, int UROWS = 4>
__global__ __launch_bounds__(kBlock) void bn_fwd_norm_kernel(
    const T* __restrict__ x, const T* __restrict__ res, T* __restrict__ y,
    long rows, long C, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, bool relu) {
  int opr, rpb, o, rl; bool active;
  tile_map(C, opr, rpb, o, rl, active);
  if (!active) return;
  const long octets = C / 8;
  for (long ob = o; ob < octets; ob += opr) {
    const long c0 = ob * 8;
    float sc[8], sh[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float g = gamma ? gamma[c0 + i] : 1.f;
      sc[i] = g * invstd[c0 + i];
      sh[i] = (beta ? beta[c0 + i] : 0.f) - mean[c0 + i] * sc[i];
    }
    const long rstride = (long)gridDim.x * rpb;
    long r = (long)blockIdx.x * rpb + rl;
    // 4x unrolled: four (eight with ADD) independent 16 B loads in
    // flight per thread — the 2x version measured only 2.6 TB/s on the
    // C=64 layers (latency-bound; the 4x-unrolled reduce kernel on the
    // same shapes sustains materially more)
    constexpr int U = UROWS;
    for (; r + (U - 1) * rstride < rows; r += U * rstride) {
      float v[U][8], a[U][8];
#pragma unroll
      for (int u = 0; u < U; ++u)
        VecIO<T>::load(x + (r + u * rstride) * C + c0, v[u]);
      if (ADD) {
#pragma unroll
        for (int u = 0; u < U; ++u)
          VecIO<T>::load(res + (r + u * rstride) * C + c0, a[u]);
      }
#pragma unroll
      for (int u = 0; u < U; ++u) {
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          float t = v[u][i] * sc[i] + sh[i];
          if (ADD) t += a[u][i];
          if (relu) t = t > 0.f ? t : 0.f;
          v[u][i] = t;
        }
      }
#pragma unroll
      for (int u = 0; u < U; ++u)
        VecIO<T>::store(y + (r + u * rstride) * C + c0, v[u]);
    }
    for (; r < rows; r += rstride) {
      float v[8], a[8];
      VecIO<T>::load(x + r * C + c0, v);
      if (ADD) VecIO<T>::load(res + r * C + c0, a);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        v[i] = v[i] * sc[i] + sh[i];
        if (ADD) v[i] += a[i];
        if (relu) v[i] = v[i] > 0.f ? v[i] : 0.f;
      }
      VecIO<T>::store(y + r * C + c0, v);
    }
  }
}

// -------------------------------------------------- bwd: reduce ---------
// RELU: the forward fused y = relu(bn(x)); backward gates dy by y>0,
// recomputing y's sign from (x, mean, invstd, gamma, beta) — no saved
// activation needed. ADD: forward was y = relu(bn(x) + res); the gate
// recomputes bn(x)+res's sign, loading res alongside x.
template <typename T, bool RELU, bool ADD>
__global__ __launch_bounds__(kBlock) void bn_bwd_reduce_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ res, long rows, long C,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ partial) {
  int opr, rpb, o, rl; bool active;
  tile_map(C, opr, rpb, o, rl, active);
  const long octets = C / 8;
  const bool wide = octets > opr;
  float sd[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float sx[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  long c0_last = 0;
  if (active) {
    for (long ob = o; ob < octets; ob += opr) {
      const long c0 = ob * 8;
      c0_last = c0;
      float mu[8], is[8], ga[8], be[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        mu[i] = mean[c0 + i];
        is[i] = invstd[c0 + i];
        if (RELU) {
          ga[i] = gamma ? gamma[c0 + i] : 1.f;
          be[i] = beta ? beta[c0 + i] : 0.f;
        }
      }
      const long rstride = (long)gridDim.x * rpb;
      long r = (long)blockIdx.x * rpb + rl;
      // U rows in flight (2U or 3U 16 B loads) — matched to the dx
      // kernel's measured sweet spot (3 rows for the 3-input case)
      constexpr int U = ADD ? 3 : 4;
      for (; r + (U - 1) * rstride < rows; r += U * rstride) {
        float g[U][8], v[U][8], a[U][8];
#pragma unroll
        for (int u = 0; u < U; ++u) {
          VecIO<T>::load(dy + (r + u * rstride) * C + c0, g[u]);
          VecIO<T>::load(x + (r + u * rstride) * C + c0, v[u]);
          if (ADD)
            VecIO<T>::load(res + (r + u * rstride) * C + c0, a[u]);
        }
#pragma unroll
        for (int u = 0; u < U; ++u) {
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            const float xh = (v[u][i] - mu[i]) * is[i];
            if (RELU) {
              const float rr = ADD ? a[u][i] : 0.f;
              if (xh * ga[i] + be[i] + rr <= 0.f) g[u][i] = 0.f;
            }
            sd[i] += g[u][i];
            sx[i] += g[u][i] * xh;
          }
        }
      }
      for (; r < rows; r += rstride) {
        float g[8], v[8], a[8];
        VecIO<T>::load(dy + r * C + c0, g);
        VecIO<T>::load(x + r * C + c0, v);
        if (ADD) VecIO<T>::load(res + r * C + c0, a);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const float xh = (v[i] - mu[i]) * is[i];
          if (RELU &&
              xh * ga[i] + be[i] + (ADD ? a[i] : 0.f) <= 0.f)
            g[i] = 0.f;
          sd[i] += g[i];
          sx[i] += g[i] * xh;
        }
      }
      if (wide) {
        float* out = partial + (long)blockIdx.x * 2 * C;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          out[c0 + i] = sd[i];
          out[C + c0 + i] = sx[i];
          sd[i] = 0; sx[i] = 0;
        }
      }
    }
  }
  if (!wide)
    block_reduce_store(sd, sx, C, opr, rpb, o, rl, active, c0_last, false,
                       partial);
}

// --------------------------------------------- bwd: finalize ------------
__global__ __launch_bounds__(kBlock) void bn_bwd_finalize_kernel(
    const float* __restrict__ partial, long nblk, long C,
    float* __restrict__ dbeta, float* __restrict__ dgamma) {
  long c; bool leader; float sd, sx;
  finalize_sums(partial, 0, nblk, C, c, leader, sd, sx);
  if (!leader) return;
  dbeta[c] = sd;
  dgamma[c] = sx;
}

// ---------------------------------------------------- bwd: dx -----------
// dx = gamma*invstd*(dy - dbeta/M - xhat*dgamma/M)
// ADD: dy is first gated by the recomputed post-add sign; the gated dy
// IS the residual branch's gradient, stored to dres in the same pass
// (saving the separate relu-backward kernel of the unfused chain).
template <typename T, bool RELU, bool ADD, int UROWS = (ADD ? 2 : 4)>
__global__ __launch_bounds__(kBlock) void bn_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ res, T* __restrict__ dx, T* __restrict__ dres,
    long rows, long C, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, const float* __restrict__ dbeta,
    const float* __restrict__ dgamma, float invM) {
  int opr, rpb, o, rl; bool active;
  tile_map(C, opr, rpb, o, rl, active);
  if (!active) return;
  const long octets = C / 8;
  for (long ob = o; ob < octets; ob += opr) {
    const long c0 = ob * 8;
    float mu[8], is[8], gi[8], md[8], mx[8], ga[8], be[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      mu[i] = mean[c0 + i];
      is[i] = invstd[c0 + i];
      ga[i] = gamma ? gamma[c0 + i] : 1.f;
      be[i] = beta ? beta[c0 + i] : 0.f;
      gi[i] = ga[i] * is[i];
      md[i] = dbeta[c0 + i] * invM;
      mx[i] = dgamma[c0 + i] * invM;
    }
    const long rstride = (long)gridDim.x * rpb;
    long r = (long)blockIdx.x * rpb + rl;
    // UROWS rows in flight: 4 for the 2-input case (dy+x = 8 loads),
    // 3 for the residual case (9 loads; +0.6% end-to-end vs 2 —
    // MGX_BN_DX_U3=0 reverts)
    constexpr int U = UROWS;
    for (; r + (U - 1) * rstride < rows; r += U * rstride) {
      float g[U][8], v[U][8], a[U][8];
#pragma unroll
      for (int u = 0; u < U; ++u) {
        VecIO<T>::load(dy + (r + u * rstride) * C + c0, g[u]);
        VecIO<T>::load(x + (r + u * rstride) * C + c0, v[u]);
        if (ADD) VecIO<T>::load(res + (r + u * rstride) * C + c0, a[u]);
      }
#pragma unroll
      for (int u = 0; u < U; ++u) {
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const float xh = (v[u][i] - mu[i]) * is[i];
          if (RELU) {
            const float rr = ADD ? a[u][i] : 0.f;
            if (xh * ga[i] + be[i] + rr <= 0.f) g[u][i] = 0.f;
          }
          if (ADD) a[u][i] = g[u][i];   // gated dy == residual grad
          g[u][i] = gi[i] * (g[u][i] - md[i] - xh * mx[i]);
        }
      }
#pragma unroll
      for (int u = 0; u < U; ++u) {
        if (ADD) VecIO<T>::store(dres + (r + u * rstride) * C + c0, a[u]);
        VecIO<T>::store(dx + (r + u * rstride) * C + c0, g[u]);
      }
    }
    for (; r < rows; r += rstride) {
      float g[8], v[8], a[8];
      VecIO<T>::load(dy + r * C + c0, g);
      VecIO<T>::load(x + r * C + c0, v);
      if (ADD) VecIO<T>::load(res + r * C + c0, a);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float xh = (v[i] - mu[i]) * is[i];
        if (RELU && xh * ga[i] + be[i] + (ADD ? a[i] : 0.f) <= 0.f)
          g[i] = 0.f;
        if (ADD) a[i] = g[i];
        g[i] = gi[i] * (g[i] - md[i] - xh * mx[i]);
      }
      if (ADD) VecIO<T>::store(dres + r * C + c0, a);
      VecIO<T>::store(dx + r * C + c0, g);
    }
  }
}

// ----------------------------------------------- NHWC max pooling -------
// Hand-written replacement for at::native::max_pool_{forward,backward}_
// nhwc: torch's NHWC maxpool backward scatters through ATOMICS and
// measured 440 us/call on VGG-16's 5 pools (11% of the step) and
// 311 us on resnet50's stem pool. Here:
//   fwd:  per output octet, scan the KxK window, store max + a uint8
//         ARGMAX (window-local 0..K*K-1) — same thread tiling as BN.
//   bwd:  GATHER per input octet: enumerate the <= ceil(K/S)^2 windows
//         that contain this input pixel, read their dy+idx octets, and
//         accumulate where the argmax points back here. No atomics, no
//         full-tensor zero-fill, deterministic.
// Gradient ties follow the saved argmax (self-consistent; torch
// likewise routes the gradient to its own saved index).

template <typename T>
__global__ __launch_bounds__(kBlock) void maxpool_fwd_kernel(
    const T* __restrict__ x, T* __restrict__ y,
    unsigned char* __restrict__ idx, long N, long C, long H, long W,
    long OH, long OW, int K, int S, int P) {
  int opr, rpb, o, rl; bool active;
  tile_map(C, opr, rpb, o, rl, active);
  if (!active) return;
  const long octets = C / 8;
  const long orows = N * OH * OW;
  const long rstride = (long)gridDim.x * rpb;
  for (long ob = o; ob < octets; ob += opr) {
    const long c0 = ob * 8;
    for (long r = (long)blockIdx.x * rpb + rl; r < orows; r += rstride) {
      const long ow = r % OW;
      const long oh = (r / OW) % OH;
      const long n = r / (OW * OH);
      const long ih0 = oh * S - P;
      const long iw0 = ow * S - P;
      float best[8];
      unsigned char bi[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        best[i] = -3.4e38f;
        bi[i] = 0;
      }
      for (int kh = 0; kh < K; ++kh) {
        const long ih = ih0 + kh;
        if (ih < 0 || ih >= H) continue;
        for (int kw = 0; kw < K; ++kw) {
          const long iw = iw0 + kw;
          if (iw < 0 || iw >= W) continue;
          float v[8];
          VecIO<T>::load(x + ((n * H + ih) * W + iw) * C + c0, v);
          const unsigned char li = (unsigned char)(kh * K + kw);
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            if (v[i] > best[i]) {
              best[i] = v[i];
              bi[i] = li;
            }
          }
        }
      }
      VecIO<T>::store(y + r * C + c0, best);
      // 8 uint8 argmaxes as one 8-byte store
      uint2 packed;
      unsigned char* pb = reinterpret_cast<unsigned char*>(&packed);
#pragma unroll
      for (int i = 0; i < 8; ++i) pb[i] = bi[i];
      *reinterpret_cast<uint2*>(idx + r * C + c0) = packed;
    }
  }
}

template <typename T>
__global__ __launch_bounds__(kBlock) void maxpool_bwd_kernel(
    const T* __restrict__ dy, const unsigned char* __restrict__ idx,
    T* __restrict__ dx, long N, long C, long H, long W, long OH, long OW,
    int K, int S, int P) {
  int opr, rpb, o, rl; bool active;
  tile_map(C, opr, rpb, o, rl, active);
  if (!active) return;
  const long octets = C / 8;
  const long irows = N * H * W;
  const long rstride = (long)gridDim.x * rpb;
  for (long ob = o; ob < octets; ob += opr) {
    const long c0 = ob * 8;
    for (long r = (long)blockIdx.x * rpb + rl; r < irows; r += rstride) {
      const long iw = r % W;
      const long ih = (r / W) % H;
      const long n = r / (W * H);
      float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      // windows (oh, ow) with oh*S - P <= ih < oh*S - P + K
      const long oh_hi = (ih + P) / S;
      const long oh_lo = (ih + P - K + S) / S;   // ceil((ih+P-K+1)/S)
      const long ow_hi = (iw + P) / S;
      const long ow_lo = (iw + P - K + S) / S;
      for (long oh = max(oh_lo, 0L); oh <= min(oh_hi, OH - 1); ++oh) {
        for (long ow = max(ow_lo, 0L); ow <= min(ow_hi, OW - 1); ++ow) {
          const long orow = (n * OH + oh) * OW + ow;
          const unsigned char li = (unsigned char)(
              (ih - (oh * S - P)) * K + (iw - (ow * S - P)));
          const uint2 packed =
              *reinterpret_cast<const uint2*>(idx + orow * C + c0);
          const unsigned char* pb =
              reinterpret_cast<const unsigned char*>(&packed);
          float g[8];
          VecIO<T>::load(dy + orow * C + c0, g);
#pragma unroll
          for (int i = 0; i < 8; ++i)
            if (pb[i] == li) acc[i] += g[i];
        }
      }
      VecIO<T>::store(dx + r * C + c0, acc);
    }
  }
}

// ------------------------------------------------- host plumbing --------

struct Geometry {
  long rows, C;
  int opr, rpb;
};

Geometry geom(const torch::Tensor& x) {
  TORCH_CHECK(x.dim() == 4, "expected 4-D NHWC tensor");
  const long N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 8 == 0, "C must be a multiple of 8");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "expected channels_last input");
  Geometry g;
  g.rows = N * H * W;
  g.C = C;
  g.opr = (int)std::min(C / 8, (long)kBlock);
  g.rpb = kBlock / g.opr;
  return g;
}

// Reduce-kernel grid: stream-filling for big inputs, but never more
// blocks than ~rows/(rpb*8) so the finalize pass over [nblk][2C]
// partials stays negligible.
int grid_reduce(const Geometry& g) {
  long nblk = (g.rows + (long)g.rpb * 8 - 1) / ((long)g.rpb * 8);
  if (nblk > 1024) nblk = 1024;  // fills HBM on the 200 MB layers; the
                                 // 8-way finalize keeps nblk=1024 cheap
  return (int)(nblk < 1 ? 1 : nblk);
}

// Elementwise-kernel grid: pure streaming, fill the chip.
int grid_elem(const Geometry& g) {
  long nblk = (g.rows + g.rpb - 1) / g.rpb;
  if (nblk > 2048) nblk = 2048;
  return (int)(nblk < 1 ? 1 : nblk);
}

constexpr int kFinSplit = 8;

// Returns the (buffer, count) the final finalize kernel should read:
// for large nblk, runs the stage1 tree first.
std::pair<float*, long> finalize_tree(torch::Tensor& partial, int rblocks,
                                      long C, hipStream_t stream) {
  float* base = partial.data_ptr<float>();
  if (rblocks <= 64) return {base, (long)rblocks};
  float* stage = base + (long)rblocks * 2 * C;
  hipLaunchKernelGGL(bn_finalize_stage1_kernel,
                     dim3((C + kFinC - 1) / kFinC, kFinSplit),
                     dim3(kBlock), 0, stream, base, (long)rblocks, C,
                     stage);
  return {stage, (long)kFinSplit};
}

#define DISPATCH_DT(scalar_type, ...)                                        \
  switch (scalar_type) {                                                     \
    case at::kFloat: {                                                       \
      using dt = float; __VA_ARGS__; break;                                  \
    }                                                                        \
    case at::kBFloat16: {                                                    \
      using dt = __hip_bfloat16; __VA_ARGS__; break;                         \
    }                                                                        \
    case at::kHalf: {                                                        \
      using dt = __half; __VA_ARGS__; break;                                 \
    }                                                                        \
    default: TORCH_CHECK(false, "unsupported dtype");                        \
  }

std::vector<torch::Tensor> bn_fwd_train(
    torch::Tensor x, torch::Tensor gamma, torch::Tensor beta,
    torch::Tensor running_mean, torch::Tensor running_var, double momentum,
    double eps, bool relu, c10::optional<torch::Tensor> residual_opt) {
  torch::Tensor residual =
      residual_opt.has_value() ? residual_opt.value() : torch::Tensor();
  auto g = geom(x);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  auto opts = torch::TensorOptions().dtype(torch::kFloat).device(x.device());
  const int rblocks = grid_reduce(g);
  const int eblocks = grid_elem(g);
  auto partial = torch::empty(
      {(long)(rblocks + (rblocks > 64 ? kFinSplit : 0)) * 2 * g.C}, opts);
  auto mean = torch::empty({g.C}, opts);
  auto invstd = torch::empty({g.C}, opts);
  auto y = torch::empty_like(x);
  const bool add = residual.defined();
  if (add) {
    TORCH_CHECK(residual.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                    residual.sizes() == x.sizes() &&
                    residual.scalar_type() == x.scalar_type(),
                "residual must match x (channels_last, shape, dtype)");
  }
  const bool fused_fin = (fused_fin_enabled() &&
                          rblocks <= kFusedFinBlocks && g.C <= 2048);
  DISPATCH_DT(x.scalar_type(), {
    hipLaunchKernelGGL(bn_fwd_reduce_kernel<dt>, dim3(rblocks),
                       dim3(kBlock), 0, stream,
                       reinterpret_cast<const dt*>(x.data_ptr()), g.rows,
                       g.C, partial.data_ptr<float>());
  });
  const float* gamma_pf = gamma.defined() ? gamma.data_ptr<float>()
                                          : nullptr;
  const float* beta_pf = beta.defined() ? beta.data_ptr<float>()
                                        : nullptr;
  float* rm_p = running_mean.defined() ? running_mean.data_ptr<float>()
                                       : nullptr;
  float* rv_p = running_var.defined() ? running_var.data_ptr<float>()
                                      : nullptr;
  if (fused_fin) {
    // finalize inlined into the normalize pass (see kFusedFinBlocks)
    DISPATCH_DT(x.scalar_type(), {
      if (add)
        hipLaunchKernelGGL((bn_fwd_norm_fin_kernel<dt, true>),
                           dim3(eblocks), dim3(kBlock), 0, stream,
                           reinterpret_cast<const dt*>(x.data_ptr()),
                           reinterpret_cast<const dt*>(
                               residual.data_ptr()),
                           reinterpret_cast<dt*>(y.data_ptr()), g.rows,
                           g.C, partial.data_ptr<float>(), (long)rblocks,
                           (float)g.rows, (float)eps, (float)momentum,
                           gamma_pf, beta_pf, mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), rm_p, rv_p, relu);
      else
        hipLaunchKernelGGL((bn_fwd_norm_fin_kernel<dt, false>),
                           dim3(eblocks), dim3(kBlock), 0, stream,
                           reinterpret_cast<const dt*>(x.data_ptr()),
                           nullptr,
                           reinterpret_cast<dt*>(y.data_ptr()), g.rows,
                           g.C, partial.data_ptr<float>(), (long)rblocks,
                           (float)g.rows, (float)eps, (float)momentum,
                           gamma_pf, beta_pf, mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), rm_p, rv_p, relu);
    });
    CHECK_HIP(hipGetLastError());
    return {y, mean, invstd};
  }
  auto fin = finalize_tree(partial, rblocks, g.C, stream);
  hipLaunchKernelGGL(bn_fwd_finalize_kernel,
                     dim3((g.C + kFinC - 1) / kFinC), dim3(kBlock), 0,
                     stream, fin.first, fin.second, g.C,
                     (float)g.rows, (float)eps, (float)momentum,
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     rm_p, rv_p);
  DISPATCH_DT(x.scalar_type(), {
    // 3 rows in flight for the residual norm (6 loads + 3 stores):
    // measured +0.3% end-to-end vs 4 rows (same trend as the dx
    // kernel: the 2-input kernels prefer 4, 3-input prefer 3);
    // MGX_BN_NORM_U3=0 reverts
    static const bool norm_u3 = [] {
      const char* v = getenv("MGX_BN_NORM_U3");
      return v == nullptr || v[0] != '0';
    }();
    if (add && norm_u3)
      hipLaunchKernelGGL((bn_fwd_norm_kernel<dt, true, 3>),
                         dim3(eblocks), dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         reinterpret_cast<const dt*>(residual.data_ptr()),
                         reinterpret_cast<dt*>(y.data_ptr()), g.rows, g.C,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma_pf, beta_pf, relu);
    else if (add)
      hipLaunchKernelGGL((bn_fwd_norm_kernel<dt, true>), dim3(eblocks),
                         dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         reinterpret_cast<const dt*>(residual.data_ptr()),
                         reinterpret_cast<dt*>(y.data_ptr()), g.rows, g.C,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma_pf, beta_pf, relu);
    else
      hipLaunchKernelGGL((bn_fwd_norm_kernel<dt, false>), dim3(eblocks),
                         dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         nullptr,
                         reinterpret_cast<dt*>(y.data_ptr()), g.rows, g.C,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma_pf, beta_pf, relu);
  });
  CHECK_HIP(hipGetLastError());
  return {y, mean, invstd};
}

torch::Tensor bn_fwd_eval(torch::Tensor x, torch::Tensor gamma,
                          torch::Tensor beta, torch::Tensor running_mean,
                          torch::Tensor running_var, double eps,
                          bool relu,
                          c10::optional<torch::Tensor> residual_opt) {
  torch::Tensor residual =
      residual_opt.has_value() ? residual_opt.value() : torch::Tensor();
  auto g = geom(x);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  auto invstd = (running_var + eps).rsqrt();
  auto y = torch::empty_like(x);
  const int eblocks = grid_elem(g);
  const bool add = residual.defined();
  DISPATCH_DT(x.scalar_type(), {
    if (add)
      hipLaunchKernelGGL((bn_fwd_norm_kernel<dt, true>), dim3(eblocks),
                         dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         reinterpret_cast<const dt*>(residual.data_ptr()),
                         reinterpret_cast<dt*>(y.data_ptr()), g.rows, g.C,
                         running_mean.data_ptr<float>(),
                         invstd.data_ptr<float>(),
                         gamma.defined() ? gamma.data_ptr<float>()
                                         : nullptr,
                         beta.defined() ? beta.data_ptr<float>() : nullptr,
                         relu);
    else
      hipLaunchKernelGGL((bn_fwd_norm_kernel<dt, false>), dim3(eblocks),
                         dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         nullptr,
                         reinterpret_cast<dt*>(y.data_ptr()), g.rows, g.C,
                         running_mean.data_ptr<float>(),
                         invstd.data_ptr<float>(),
                         gamma.defined() ? gamma.data_ptr<float>()
                                         : nullptr,
                         beta.defined() ? beta.data_ptr<float>() : nullptr,
                         relu);
  });
  CHECK_HIP(hipGetLastError());
  return y;
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor mean, torch::Tensor invstd,
                                  torch::Tensor gamma, torch::Tensor beta,
                                  bool relu,
                                  c10::optional<torch::Tensor> residual_opt) {
  torch::Tensor residual =
      residual_opt.has_value() ? residual_opt.value() : torch::Tensor();
  auto g = geom(x);
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast),
              "expected channels_last grad");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  auto opts = torch::TensorOptions().dtype(torch::kFloat).device(x.device());
  const int rblocks = grid_reduce(g);
  const int eblocks = grid_elem(g);
  auto partial = torch::empty(
      {(long)(rblocks + (rblocks > 64 ? kFinSplit : 0)) * 2 * g.C}, opts);
  auto dbeta = torch::empty({g.C}, opts);
  auto dgamma = torch::empty({g.C}, opts);
  auto dx = torch::empty_like(x);
  const bool add = residual.defined();
  TORCH_CHECK(!add || relu, "residual fusion implies post-add relu");
  torch::Tensor dres;
  const float* gamma_p = gamma.defined() ? gamma.data_ptr<float>()
                                         : nullptr;
  const float* beta_p = beta.defined() ? beta.data_ptr<float>() : nullptr;
  DISPATCH_DT(x.scalar_type(), {
    if (add)
      hipLaunchKernelGGL((bn_bwd_reduce_kernel<dt, true, true>),
                         dim3(rblocks), dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(dy.data_ptr()),
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         reinterpret_cast<const dt*>(residual.data_ptr()),
                         g.rows, g.C, mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), gamma_p, beta_p,
                         partial.data_ptr<float>());
    else if (relu)
      hipLaunchKernelGGL((bn_bwd_reduce_kernel<dt, true, false>),
                         dim3(rblocks), dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(dy.data_ptr()),
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         nullptr, g.rows,
                         g.C, mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), gamma_p, beta_p,
                         partial.data_ptr<float>());
    else
      hipLaunchKernelGGL((bn_bwd_reduce_kernel<dt, false, false>),
                         dim3(rblocks), dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(dy.data_ptr()),
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         nullptr, g.rows,
                         g.C, mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), gamma_p, beta_p,
                         partial.data_ptr<float>());
  });
  const bool fused_fin = (fused_fin_enabled() &&
                          rblocks <= kFusedFinBlocks && g.C <= 2048);
  if (fused_fin) {
    DISPATCH_DT(x.scalar_type(), {
      const dt* res_p = add
          ? reinterpret_cast<const dt*>(residual.data_ptr()) : nullptr;
      dt* dres_p = nullptr;
      if (add) {
        dres = torch::empty_like(x);
        dres_p = reinterpret_cast<dt*>(dres.data_ptr());
      }
      auto launch = [&](auto kern) {
        hipLaunchKernelGGL(kern, dim3(eblocks), dim3(kBlock), 0, stream,
                           reinterpret_cast<const dt*>(dy.data_ptr()),
                           reinterpret_cast<const dt*>(x.data_ptr()),
                           res_p, reinterpret_cast<dt*>(dx.data_ptr()),
                           dres_p, g.rows, g.C, mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), gamma_p, beta_p,
                           partial.data_ptr<float>(), (long)rblocks,
                           1.f / (float)g.rows, dbeta.data_ptr<float>(),
                           dgamma.data_ptr<float>());
      };
      if (add)
        launch(bn_bwd_dx_fin_kernel<dt, true, true>);
      else if (relu)
        launch(bn_bwd_dx_fin_kernel<dt, true, false>);
      else
        launch(bn_bwd_dx_fin_kernel<dt, false, false>);
    });
    CHECK_HIP(hipGetLastError());
    if (add) return {dx, dgamma, dbeta, dres};
    return {dx, dgamma, dbeta};
  }
  auto finb = finalize_tree(partial, rblocks, g.C, stream);
  hipLaunchKernelGGL(bn_bwd_finalize_kernel,
                     dim3((g.C + kFinC - 1) / kFinC), dim3(kBlock), 0,
                     stream, finb.first, finb.second, g.C,
                     dbeta.data_ptr<float>(), dgamma.data_ptr<float>());
  // 3 rows in flight for the residual dx (9 loads): measured +0.6%
  // end-to-end on resnet50 vs 2 rows (same-box repeated A/B);
  // MGX_BN_DX_U3=0 reverts
  static const bool dx_u3 = [] {
    const char* v = getenv("MGX_BN_DX_U3");
    return v == nullptr || v[0] != '0';
  }();
  DISPATCH_DT(x.scalar_type(), {
    if (add && dx_u3) {
      dres = torch::empty_like(x);
      hipLaunchKernelGGL((bn_bwd_dx_kernel<dt, true, true, 3>),
                         dim3(eblocks), dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(dy.data_ptr()),
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         reinterpret_cast<const dt*>(residual.data_ptr()),
                         reinterpret_cast<dt*>(dx.data_ptr()),
                         reinterpret_cast<dt*>(dres.data_ptr()),
                         g.rows, g.C,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma_p, beta_p, dbeta.data_ptr<float>(),
                         dgamma.data_ptr<float>(), 1.f / (float)g.rows);
    } else if (add) {
      dres = torch::empty_like(x);
      hipLaunchKernelGGL((bn_bwd_dx_kernel<dt, true, true>),
                         dim3(eblocks), dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(dy.data_ptr()),
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         reinterpret_cast<const dt*>(residual.data_ptr()),
                         reinterpret_cast<dt*>(dx.data_ptr()),
                         reinterpret_cast<dt*>(dres.data_ptr()),
                         g.rows, g.C,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma_p, beta_p, dbeta.data_ptr<float>(),
                         dgamma.data_ptr<float>(), 1.f / (float)g.rows);
    } else if (relu)
      hipLaunchKernelGGL((bn_bwd_dx_kernel<dt, true, false>),
                         dim3(eblocks), dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(dy.data_ptr()),
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         nullptr,
                         reinterpret_cast<dt*>(dx.data_ptr()), nullptr,
                         g.rows, g.C,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma_p, beta_p, dbeta.data_ptr<float>(),
                         dgamma.data_ptr<float>(), 1.f / (float)g.rows);
    else
      hipLaunchKernelGGL((bn_bwd_dx_kernel<dt, false, false>),
                         dim3(eblocks), dim3(kBlock), 0, stream,
                         reinterpret_cast<const dt*>(dy.data_ptr()),
                         reinterpret_cast<const dt*>(x.data_ptr()),
                         nullptr,
                         reinterpret_cast<dt*>(dx.data_ptr()), nullptr,
                         g.rows, g.C,
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma_p, beta_p, dbeta.data_ptr<float>(),
                         dgamma.data_ptr<float>(), 1.f / (float)g.rows);
  });
  CHECK_HIP(hipGetLastError());
  if (add) return {dx, dgamma, dbeta, dres};
  return {dx, dgamma, dbeta};
}

std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x, long K, long S,
                                       long P) {
  auto g = geom(x);
  const long N = x.size(0), H = x.size(2), W = x.size(3);
  const long OH = (H + 2 * P - K) / S + 1;
  const long OW = (W + 2 * P - K) / S + 1;
  TORCH_CHECK(K * K <= 255, "argmax stored as uint8");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  auto y = torch::empty(
      {N, g.C, OH, OW},
      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto idx = torch::empty(
      {N * OH * OW * g.C},
      torch::TensorOptions().dtype(torch::kUInt8).device(x.device()));
  const long orows = N * OH * OW;
  long nblk = (orows + g.rpb - 1) / g.rpb;
  if (nblk > 2048) nblk = 2048;
  DISPATCH_DT(x.scalar_type(), {
    hipLaunchKernelGGL(maxpool_fwd_kernel<dt>,
                       dim3((int)std::max(nblk, 1L)), dim3(kBlock), 0,
                       stream, reinterpret_cast<const dt*>(x.data_ptr()),
                       reinterpret_cast<dt*>(y.data_ptr()),
                       idx.data_ptr<unsigned char>(), N, g.C, H, W, OH,
                       OW, (int)K, (int)S, (int)P);
  });
  CHECK_HIP(hipGetLastError());
  return {y, idx};
}

torch::Tensor maxpool_bwd(torch::Tensor dy, torch::Tensor idx, long N,
                          long C, long H, long W, long K, long S, long P) {
  auto gdy = geom(dy);
  const long OH = dy.size(2), OW = dy.size(3);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  auto dx = torch::empty(
      {N, C, H, W},
      dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  const long irows = N * H * W;
  const int opr = (int)std::min(C / 8, (long)kBlock);
  const int rpb = kBlock / opr;
  long nblk = (irows + rpb - 1) / rpb;
  if (nblk > 2048) nblk = 2048;
  DISPATCH_DT(dy.scalar_type(), {
    hipLaunchKernelGGL(maxpool_bwd_kernel<dt>,
                       dim3((int)std::max(nblk, 1L)), dim3(kBlock), 0,
                       stream,
                       reinterpret_cast<const dt*>(dy.data_ptr()),
                       idx.data_ptr<unsigned char>(),
                       reinterpret_cast<dt*>(dx.data_ptr()), N, C, H, W,
                       OH, OW, (int)K, (int)S, (int)P);
  });
  CHECK_HIP(hipGetLastError());
  (void)gdy;
  return dx;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("maxpool_fwd", &maxpool_fwd,
        "NHWC max pool forward; returns (y, uint8 argmax)");
  m.def("maxpool_bwd", &maxpool_bwd,
        "NHWC max pool backward (gather, no atomics)");
  m.def("bn_fwd_train", &bn_fwd_train,
        "NHWC BN training forward (optionally y=relu(bn(x)+residual)); "
        "returns (y, mean, invstd)",
        py::arg("x"), py::arg("gamma"), py::arg("beta"),
        py::arg("running_mean"), py::arg("running_var"),
        py::arg("momentum"), py::arg("eps"), py::arg("relu"),
        py::arg("residual") = py::none());
  m.def("bn_fwd_eval", &bn_fwd_eval, "NHWC BN inference forward",
        py::arg("x"), py::arg("gamma"), py::arg("beta"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("eps"),
        py::arg("relu"), py::arg("residual") = py::none());
  m.def("bn_bwd", &bn_bwd,
        "NHWC BN backward; returns (dx, dgamma, dbeta[, dresidual])",
        py::arg("dy"), py::arg("x"), py::arg("mean"), py::arg("invstd"),
        py::arg("gamma"), py::arg("beta"), py::arg("relu"),
        py::arg("residual") = py::none());
}
