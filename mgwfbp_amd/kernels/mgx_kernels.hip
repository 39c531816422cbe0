// MI355X (gfx950/CDNA4) hot-path kernels for merged-gradient WFBP training.
//
// Native replacements for the per-layer torch ops the reference delegates
// to its framework (reference distributed_optimizer.py:311 pack copies,
// :327-331 unpack views, :383-389 norm/scale, and torch.optim.SGD's many
// small per-tensor kernels at dl_trainer.py:244-248):
//
//   multi_tensor_sgd     — fused SGD + momentum + weight-decay (+nesterov)
//                          over ALL parameters in ONE launch
//   multi_tensor_pack    — gather per-layer grads into a flat comm buffer
//                          with optional fp32->bf16/fp16 cast + scale
//   multi_tensor_unpack  — scatter flat buffer back with cast + scale
//   l2norm               — squared-L2 reduction over a tensor list
//   scale_inplace        — flat buffer *= s (clip / average epilogue)
//
// Design (cdna_hip_programming.md Appendix B "Element-wise", G11, G13):
// all kernels are HBM-bandwidth-bound streaming ops. 256-thread blocks
// (4 waves of 64), 16 B/lane vectorized access (float4 / 8xbf16), grid =
// min(chunks, 2048) with grid-stride over a device-resident chunk table
// (built once per tensor-list and cached host-side; tensor addresses are
// stable across steps so steady state launches touch no host memory).
// Per-wave shuffle reduction + one atomicAdd per block for the norm (G12).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <vector>

#define CHECK_HIP(x)                                                         \
  do {                                                                       \
    hipError_t err__ = (x);                                                  \
    TORCH_CHECK(err__ == hipSuccess, "HIP error: ",                          \
                hipGetErrorString(err__));                                   \
  } while (0)

namespace {

constexpr int kBlock = 256;          // 4 wave64s
constexpr long kChunk = 1 << 16;     // elements per chunk (64K)
constexpr int kMaxGrid = 2048;       // 256 CUs * 8 blocks

struct TensorTriple {
  float* p;        // param
  float* g;        // grad
  float* m;        // momentum buffer (nullptr if momentum==0)
  long numel;
  float wd;        // per-tensor weight decay (bn/bias no-decay split)
};

struct PackDesc {
  const void* src;   // per-layer grad
  long offset;       // element offset in flat buffer
  long numel;
};

struct Chunk {
  int tensor_idx;
  long start;        // element offset within the tensor
};

// ---------------------------------------------------------------- SGD ----
// v = mu*v + (g + wd*p); p -= lr * (nesterov ? g + wd*p + mu*v : v)
// Momentum buffers are zero-initialized host-side, so the first step's
// v = mu*0 + d_p == d_p matches torch.optim.SGD's clone-on-first-step.
// lr_ptr: optional device-resident learning rate — lets a hipGraph
// capture of the step keep a changing LR schedule without re-capture
// (the host fills the buffer before each replay).
__global__ __launch_bounds__(kBlock) void multi_tensor_sgd_kernel(
    const Chunk* __restrict__ chunks, int nchunks,
    const TensorTriple* __restrict__ tensors, float lr,
    const float* __restrict__ lr_ptr, float momentum,
    float dampening, bool nesterov, float grad_scale) {
  if (lr_ptr != nullptr) lr = *lr_ptr;
  for (int c = blockIdx.x; c < nchunks; c += gridDim.x) {
    const Chunk ck = chunks[c];
    const TensorTriple t = tensors[ck.tensor_idx];
    const long n = min(kChunk, t.numel - ck.start);
    float* __restrict__ p = t.p + ck.start;
    float* __restrict__ g = t.g + ck.start;
    float* __restrict__ m = t.m ? t.m + ck.start : nullptr;
    const float wd = t.wd;
    const long nvec = n & ~3L;  // bases are 16B-aligned; vectorize in 4s
    // 2x unrolled main loop: 6-8 independent 16B loads in flight per
    // thread (single-slot loops leave the kernel latency-bound)
    const long step = (long)blockDim.x * 4L;
    long i = threadIdx.x * 4L;
    for (; i + step < nvec; i += 2 * step) {
      float4 pv0 = *reinterpret_cast<float4*>(p + i);
      float4 gv0 = *reinterpret_cast<float4*>(g + i);
      float4 pv1 = *reinterpret_cast<float4*>(p + i + step);
      float4 gv1 = *reinterpret_cast<float4*>(g + i + step);
      float4 mv0, mv1;
      if (m) {
        mv0 = *reinterpret_cast<float4*>(m + i);
        mv1 = *reinterpret_cast<float4*>(m + i + step);
      }
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        float4& pv = u ? pv1 : pv0;
        const float4& gv = u ? gv1 : gv0;
        float4& mv = u ? mv1 : mv0;
        float dp0 = gv.x * grad_scale + wd * pv.x;
        float dp1 = gv.y * grad_scale + wd * pv.y;
        float dp2 = gv.z * grad_scale + wd * pv.z;
        float dp3 = gv.w * grad_scale + wd * pv.w;
        if (m) {
          mv.x = momentum * mv.x + (1.f - dampening) * dp0;
          mv.y = momentum * mv.y + (1.f - dampening) * dp1;
          mv.z = momentum * mv.z + (1.f - dampening) * dp2;
          mv.w = momentum * mv.w + (1.f - dampening) * dp3;
          if (nesterov) {
            dp0 += momentum * mv.x; dp1 += momentum * mv.y;
            dp2 += momentum * mv.z; dp3 += momentum * mv.w;
          } else {
            dp0 = mv.x; dp1 = mv.y; dp2 = mv.z; dp3 = mv.w;
          }
        }
        pv.x -= lr * dp0; pv.y -= lr * dp1;
        pv.z -= lr * dp2; pv.w -= lr * dp3;
      }
      if (m) {
        *reinterpret_cast<float4*>(m + i) = mv0;
        *reinterpret_cast<float4*>(m + i + step) = mv1;
      }
      *reinterpret_cast<float4*>(p + i) = pv0;
      *reinterpret_cast<float4*>(p + i + step) = pv1;
    }
    for (; i < nvec; i += step) {
      float4 pv = *reinterpret_cast<float4*>(p + i);
      float4 gv = *reinterpret_cast<float4*>(g + i);
      float dp0 = gv.x * grad_scale + wd * pv.x;
      float dp1 = gv.y * grad_scale + wd * pv.y;
      float dp2 = gv.z * grad_scale + wd * pv.z;
      float dp3 = gv.w * grad_scale + wd * pv.w;
      if (m) {
        float4 mv = *reinterpret_cast<float4*>(m + i);
        mv.x = momentum * mv.x + (1.f - dampening) * dp0;
        mv.y = momentum * mv.y + (1.f - dampening) * dp1;
        mv.z = momentum * mv.z + (1.f - dampening) * dp2;
        mv.w = momentum * mv.w + (1.f - dampening) * dp3;
        *reinterpret_cast<float4*>(m + i) = mv;
        if (nesterov) {
          dp0 += momentum * mv.x; dp1 += momentum * mv.y;
          dp2 += momentum * mv.z; dp3 += momentum * mv.w;
        } else {
          dp0 = mv.x; dp1 = mv.y; dp2 = mv.z; dp3 = mv.w;
        }
      }
      pv.x -= lr * dp0; pv.y -= lr * dp1;
      pv.z -= lr * dp2; pv.w -= lr * dp3;
      *reinterpret_cast<float4*>(p + i) = pv;
    }
    for (long i = nvec + threadIdx.x; i < n; i += blockDim.x) {
      float dp = g[i] * grad_scale + wd * p[i];
      if (m) {
        float mv = momentum * m[i] + (1.f - dampening) * dp;
        m[i] = mv;
        dp = nesterov ? dp + momentum * mv : mv;
      }
      p[i] -= lr * dp;
    }
  }
}

// ------------------------------------------------------------- pack ------
// Gather per-layer grads into the flat comm buffer. DTYPE: 0=f32, 1=bf16,
// 2=f16 (OUT dtype for pack, IN dtype for unpack; the per-layer side is
// always f32 master grads).
template <int DTYPE>
__global__ __launch_bounds__(kBlock) void multi_tensor_pack_kernel(
    const Chunk* __restrict__ chunks, int nchunks,
    const PackDesc* __restrict__ descs, void* __restrict__ dst_raw,
    float scale) {
  for (int c = blockIdx.x; c < nchunks; c += gridDim.x) {
    const Chunk ck = chunks[c];
    const PackDesc d = descs[ck.tensor_idx];
    const long n = min(kChunk, d.numel - ck.start);
    const float* __restrict__ src =
        reinterpret_cast<const float*>(d.src) + ck.start;
    const long dst_off = d.offset + ck.start;
    // flat-buffer offsets are padded to 64-element boundaries host-side,
    // so dst_off is 16B-aligned for every dtype here.
    const long nvec = n & ~3L;
    if constexpr (DTYPE == 0) {
      float* __restrict__ dst = reinterpret_cast<float*>(dst_raw) + dst_off;
      for (long i = threadIdx.x * 4L; i < nvec; i += (long)blockDim.x * 4L) {
        float4 v = *reinterpret_cast<const float4*>(src + i);
        v.x *= scale; v.y *= scale; v.z *= scale; v.w *= scale;
        *reinterpret_cast<float4*>(dst + i) = v;
      }
      for (long i = nvec + threadIdx.x; i < n; i += blockDim.x)
        dst[i] = src[i] * scale;
    } else if constexpr (DTYPE == 1) {
      __hip_bfloat16* __restrict__ dst =
          reinterpret_cast<__hip_bfloat16*>(dst_raw) + dst_off;
      for (long i = threadIdx.x * 4L; i < nvec; i += (long)blockDim.x * 4L) {
        float4 v = *reinterpret_cast<const float4*>(src + i);
        __hip_bfloat162 lo = {__float2bfloat16(v.x * scale),
                              __float2bfloat16(v.y * scale)};
        __hip_bfloat162 hi = {__float2bfloat16(v.z * scale),
                              __float2bfloat16(v.w * scale)};
        *reinterpret_cast<__hip_bfloat162*>(dst + i) = lo;
        *reinterpret_cast<__hip_bfloat162*>(dst + i + 2) = hi;
      }
      for (long i = nvec + threadIdx.x; i < n; i += blockDim.x)
        dst[i] = __float2bfloat16(src[i] * scale);
    } else {
      __half* __restrict__ dst = reinterpret_cast<__half*>(dst_raw) + dst_off;
      for (long i = threadIdx.x * 4L; i < nvec; i += (long)blockDim.x * 4L) {
        float4 v = *reinterpret_cast<const float4*>(src + i);
        __half2 lo = __floats2half2_rn(v.x * scale, v.y * scale);
        __half2 hi = __floats2half2_rn(v.z * scale, v.w * scale);
        *reinterpret_cast<__half2*>(dst + i) = lo;
        *reinterpret_cast<__half2*>(dst + i + 2) = hi;
      }
      for (long i = nvec + threadIdx.x; i < n; i += blockDim.x)
        dst[i] = __float2half(src[i] * scale);
    }
  }
}

// ------------------------------------------------------------ unpack -----
template <int DTYPE>
__global__ __launch_bounds__(kBlock) void multi_tensor_unpack_kernel(
    const Chunk* __restrict__ chunks, int nchunks,
    const PackDesc* __restrict__ descs, const void* __restrict__ src_raw,
    float scale) {
  for (int c = blockIdx.x; c < nchunks; c += gridDim.x) {
    const Chunk ck = chunks[c];
    const PackDesc d = descs[ck.tensor_idx];
    const long n = min(kChunk, d.numel - ck.start);
    float* __restrict__ dst =
        reinterpret_cast<float*>(const_cast<void*>(d.src)) + ck.start;
    const long src_off = d.offset + ck.start;
    const long nvec = n & ~3L;
    if constexpr (DTYPE == 0) {
      const float* __restrict__ src =
          reinterpret_cast<const float*>(src_raw) + src_off;
      for (long i = threadIdx.x * 4L; i < nvec; i += (long)blockDim.x * 4L) {
        float4 v = *reinterpret_cast<const float4*>(src + i);
        v.x *= scale; v.y *= scale; v.z *= scale; v.w *= scale;
        *reinterpret_cast<float4*>(dst + i) = v;
      }
      for (long i = nvec + threadIdx.x; i < n; i += blockDim.x)
        dst[i] = src[i] * scale;
    } else if constexpr (DTYPE == 1) {
      const __hip_bfloat16* __restrict__ src =
          reinterpret_cast<const __hip_bfloat16*>(src_raw) + src_off;
      for (long i = threadIdx.x * 4L; i < nvec; i += (long)blockDim.x * 4L) {
        __hip_bfloat162 lo = *reinterpret_cast<const __hip_bfloat162*>(src + i);
        __hip_bfloat162 hi =
            *reinterpret_cast<const __hip_bfloat162*>(src + i + 2);
        float4 v = {__bfloat162float(lo.x) * scale,
                    __bfloat162float(lo.y) * scale,
                    __bfloat162float(hi.x) * scale,
                    __bfloat162float(hi.y) * scale};
        *reinterpret_cast<float4*>(dst + i) = v;
      }
      for (long i = nvec + threadIdx.x; i < n; i += blockDim.x)
        dst[i] = __bfloat162float(src[i]) * scale;
    } else {
      const __half* __restrict__ src =
          reinterpret_cast<const __half*>(src_raw) + src_off;
      for (long i = threadIdx.x * 4L; i < nvec; i += (long)blockDim.x * 4L) {
        __half2 lo = *reinterpret_cast<const __half2*>(src + i);
        __half2 hi = *reinterpret_cast<const __half2*>(src + i + 2);
        float4 v = {__half2float(lo.x) * scale, __half2float(lo.y) * scale,
                    __half2float(hi.x) * scale, __half2float(hi.y) * scale};
        *reinterpret_cast<float4*>(dst + i) = v;
      }
      for (long i = nvec + threadIdx.x; i < n; i += blockDim.x)
        dst[i] = __half2float(src[i]) * scale;
    }
  }
}

// ------------------------------------------------------------- norm ------
__global__ __launch_bounds__(kBlock) void l2norm_sq_kernel(
    const Chunk* __restrict__ chunks, int nchunks,
    const PackDesc* __restrict__ descs, float* __restrict__ out) {
  float acc = 0.f;
  for (int c = blockIdx.x; c < nchunks; c += gridDim.x) {
    const Chunk ck = chunks[c];
    const PackDesc d = descs[ck.tensor_idx];
    const long n = min(kChunk, d.numel - ck.start);
    const float* __restrict__ src =
        reinterpret_cast<const float*>(d.src) + ck.start;
    const long nvec = n & ~3L;
    for (long i = threadIdx.x * 4L; i < nvec; i += (long)blockDim.x * 4L) {
      float4 v = *reinterpret_cast<const float4*>(src + i);
      acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
    for (long i = nvec + threadIdx.x; i < n; i += blockDim.x)
      acc += src[i] * src[i];
  }
  // wave64 shuffle reduce, then LDS across the 4 waves, one atomic/block
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, 64);
  __shared__ float warp_sums[kBlock / 64];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  if (lane == 0) warp_sums[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < kBlock / 64; ++w) s += warp_sums[w];
    atomicAdd(out, s);
  }
}

// Squared-L2 of a flat fp32 buffer (grid-stride, wave64 shuffle reduce,
// one atomicAdd per block). Feeds clip_scale_kernel: together they form
// the device-side gradient clip (reference distributed_optimizer.py:
// 380-389 does norm+clip through host .item() round-trips).
__global__ __launch_bounds__(kBlock) void l2norm_sq_flat_kernel(
    const float* __restrict__ buf, long n, float* __restrict__ out) {
  float acc = 0.f;
  const long nvec = n & ~3L;
  const long stride = (long)gridDim.x * blockDim.x * 4L;
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4L; i < nvec;
       i += stride) {
    float4 v = *reinterpret_cast<const float4*>(buf + i);
    acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
  }
  const long base = nvec + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (base < n) acc += buf[base] * buf[base];
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, 64);
  __shared__ float warp_sums[kBlock / 64];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  if (lane == 0) warp_sums[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < kBlock / 64; ++w) s += warp_sums[w];
    atomicAdd(out, s);
  }
}

// buf *= min(1, max_norm / (sqrt(normsq[0]) + eps)) — the clip
// coefficient is computed per-thread from the device-resident norm, so
// the whole clip is two launches with zero host round-trips (and is
// hipGraph-capturable).
__global__ __launch_bounds__(kBlock) void clip_scale_kernel(
    float* __restrict__ buf, long n, const float* __restrict__ normsq,
    float max_norm, float eps) {
  const float coef =
      fminf(1.f, max_norm / (sqrtf(normsq[0]) + eps));
  if (coef >= 1.f) return;
  const long nvec = n & ~3L;
  const long stride = (long)gridDim.x * blockDim.x * 4L;
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4L; i < nvec;
       i += stride) {
    float4 v = *reinterpret_cast<float4*>(buf + i);
    v.x *= coef; v.y *= coef; v.z *= coef; v.w *= coef;
    *reinterpret_cast<float4*>(buf + i) = v;
  }
  const long base = nvec + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (base < n) buf[base] *= coef;
}

__global__ __launch_bounds__(kBlock) void scale_inplace_kernel(
    float* __restrict__ buf, long n, float scale) {
  const long nvec = n & ~3L;
  const long stride = (long)gridDim.x * blockDim.x * 4L;
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4L; i < nvec;
       i += stride) {
    float4 v = *reinterpret_cast<float4*>(buf + i);
    v.x *= scale; v.y *= scale; v.z *= scale; v.w *= scale;
    *reinterpret_cast<float4*>(buf + i) = v;
  }
  const long base = nvec + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (base < n) buf[base] *= scale;
}

// ------------------------------------------------------- host plumbing ---
// Chunk/descriptor tables live in device memory, built once per tensor
// list and cached by the Python wrapper (addresses are stable).

int grid_for(int nchunks) { return std::min(nchunks, kMaxGrid); }

std::vector<torch::Tensor> build_sgd_table(
    const std::vector<torch::Tensor>& params,
    const std::vector<torch::Tensor>& grads,
    const std::vector<torch::Tensor>& momenta,
    const std::vector<double>& wds) {
  const int nt = params.size();
  TORCH_CHECK(nt > 0, "empty tensor list");
  std::vector<TensorTriple> triples(nt);
  std::vector<Chunk> chunks;
  for (int i = 0; i < nt; ++i) {
    TORCH_CHECK(params[i].is_contiguous() && grads[i].is_contiguous(),
                "params/grads must be contiguous");
    TORCH_CHECK(params[i].scalar_type() == at::kFloat,
                "fused SGD expects fp32 master params");
    triples[i].p = params[i].data_ptr<float>();
    triples[i].g = grads[i].data_ptr<float>();
    triples[i].m =
        momenta.empty() ? nullptr : momenta[i].data_ptr<float>();
    triples[i].numel = params[i].numel();
    triples[i].wd = (float)wds[i];
    for (long s = 0; s < triples[i].numel; s += kChunk)
      chunks.push_back({i, s});
  }
  auto opts = torch::TensorOptions()
                  .dtype(torch::kUInt8)
                  .device(params[0].device());
  auto t_tensors = torch::empty({(long)(nt * sizeof(TensorTriple))}, opts);
  auto t_chunks =
      torch::empty({(long)(chunks.size() * sizeof(Chunk))}, opts);
  CHECK_HIP(hipMemcpyAsync(t_tensors.data_ptr(), triples.data(),
                           nt * sizeof(TensorTriple), hipMemcpyHostToDevice,
                           c10::hip::getCurrentHIPStream().stream()));
  CHECK_HIP(hipMemcpyAsync(t_chunks.data_ptr(), chunks.data(),
                           chunks.size() * sizeof(Chunk),
                           hipMemcpyHostToDevice,
                           c10::hip::getCurrentHIPStream().stream()));
  CHECK_HIP(hipStreamSynchronize(c10::hip::getCurrentHIPStream().stream()));
  auto t_n = torch::tensor({(long)chunks.size()}, torch::kLong);
  return {t_tensors, t_chunks, t_n};
}

void multi_tensor_sgd(torch::Tensor t_tensors, torch::Tensor t_chunks,
                      long nchunks, double lr, torch::Tensor lr_buf,
                      double momentum, double dampening, bool nesterov,
                      double grad_scale) {
  const float* lr_ptr =
      lr_buf.defined() && lr_buf.numel() > 0
          ? lr_buf.data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(multi_tensor_sgd_kernel, dim3(grid_for(nchunks)),
                     dim3(kBlock), 0, c10::hip::getCurrentHIPStream().stream(),
                     reinterpret_cast<const Chunk*>(t_chunks.data_ptr()),
                     (int)nchunks,
                     reinterpret_cast<const TensorTriple*>(
                         t_tensors.data_ptr()),
                     (float)lr, lr_ptr, (float)momentum, (float)dampening,
                     nesterov, (float)grad_scale);
  CHECK_HIP(hipGetLastError());
}

std::vector<torch::Tensor> build_pack_table(
    const std::vector<torch::Tensor>& srcs,
    const std::vector<long>& offsets) {
  const int nt = srcs.size();
  TORCH_CHECK(nt > 0 && (int)offsets.size() == nt, "bad pack table args");
  std::vector<PackDesc> descs(nt);
  std::vector<Chunk> chunks;
  for (int i = 0; i < nt; ++i) {
    TORCH_CHECK(srcs[i].is_contiguous(), "pack sources must be contiguous");
    descs[i].src = srcs[i].data_ptr();
    descs[i].offset = offsets[i];
    descs[i].numel = srcs[i].numel();
    for (long s = 0; s < descs[i].numel; s += kChunk)
      chunks.push_back({i, s});
  }
  auto opts =
      torch::TensorOptions().dtype(torch::kUInt8).device(srcs[0].device());
  auto t_descs = torch::empty({(long)(nt * sizeof(PackDesc))}, opts);
  auto t_chunks =
      torch::empty({(long)(chunks.size() * sizeof(Chunk))}, opts);
  CHECK_HIP(hipMemcpyAsync(t_descs.data_ptr(), descs.data(),
                           nt * sizeof(PackDesc), hipMemcpyHostToDevice,
                           c10::hip::getCurrentHIPStream().stream()));
  CHECK_HIP(hipMemcpyAsync(t_chunks.data_ptr(), chunks.data(),
                           chunks.size() * sizeof(Chunk),
                           hipMemcpyHostToDevice,
                           c10::hip::getCurrentHIPStream().stream()));
  CHECK_HIP(hipStreamSynchronize(c10::hip::getCurrentHIPStream().stream()));
  auto t_n = torch::tensor({(long)chunks.size()}, torch::kLong);
  return {t_descs, t_chunks, t_n};
}

static int dtype_code(at::ScalarType t) {
  switch (t) {
    case at::kFloat: return 0;
    case at::kBFloat16: return 1;
    case at::kHalf: return 2;
    default: TORCH_CHECK(false, "unsupported comm dtype");
  }
}

void multi_tensor_pack(torch::Tensor t_descs, torch::Tensor t_chunks,
                       long nchunks, torch::Tensor flat, double scale) {
  const int code = dtype_code(flat.scalar_type());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  auto* chunks = reinterpret_cast<const Chunk*>(t_chunks.data_ptr());
  auto* descs = reinterpret_cast<const PackDesc*>(t_descs.data_ptr());
  const int grid = grid_for(nchunks);
  switch (code) {
    case 0:
      hipLaunchKernelGGL(multi_tensor_pack_kernel<0>, dim3(grid),
                         dim3(kBlock), 0, stream, chunks, (int)nchunks,
                         descs, flat.data_ptr(), (float)scale);
      break;
    case 1:
      hipLaunchKernelGGL(multi_tensor_pack_kernel<1>, dim3(grid),
                         dim3(kBlock), 0, stream, chunks, (int)nchunks,
                         descs, flat.data_ptr(), (float)scale);
      break;
    default:
      hipLaunchKernelGGL(multi_tensor_pack_kernel<2>, dim3(grid),
                         dim3(kBlock), 0, stream, chunks, (int)nchunks,
                         descs, flat.data_ptr(), (float)scale);
  }
  CHECK_HIP(hipGetLastError());
}

void multi_tensor_unpack(torch::Tensor t_descs, torch::Tensor t_chunks,
                         long nchunks, torch::Tensor flat, double scale) {
  const int code = dtype_code(flat.scalar_type());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  auto* chunks = reinterpret_cast<const Chunk*>(t_chunks.data_ptr());
  auto* descs = reinterpret_cast<const PackDesc*>(t_descs.data_ptr());
  const int grid = grid_for(nchunks);
  switch (code) {
    case 0:
      hipLaunchKernelGGL(multi_tensor_unpack_kernel<0>, dim3(grid),
                         dim3(kBlock), 0, stream, chunks, (int)nchunks,
                         descs, flat.data_ptr(), (float)scale);
      break;
    case 1:
      hipLaunchKernelGGL(multi_tensor_unpack_kernel<1>, dim3(grid),
                         dim3(kBlock), 0, stream, chunks, (int)nchunks,
                         descs, flat.data_ptr(), (float)scale);
      break;
    default:
      hipLaunchKernelGGL(multi_tensor_unpack_kernel<2>, dim3(grid),
                         dim3(kBlock), 0, stream, chunks, (int)nchunks,
                         descs, flat.data_ptr(), (float)scale);
  }
  CHECK_HIP(hipGetLastError());
}

torch::Tensor l2norm_sq(torch::Tensor t_descs, torch::Tensor t_chunks,
                        long nchunks) {
  auto out = torch::zeros(
      {1}, torch::TensorOptions().dtype(torch::kFloat).device(
               t_descs.device()));
  hipLaunchKernelGGL(l2norm_sq_kernel, dim3(grid_for(nchunks)), dim3(kBlock),
                     0, c10::hip::getCurrentHIPStream().stream(),
                     reinterpret_cast<const Chunk*>(t_chunks.data_ptr()),
                     (int)nchunks,
                     reinterpret_cast<const PackDesc*>(t_descs.data_ptr()),
                     out.data_ptr<float>());
  CHECK_HIP(hipGetLastError());
  return out;
}

void l2norm_clip_(torch::Tensor buf, torch::Tensor normsq_scratch,
                  double max_norm, double eps) {
  TORCH_CHECK(buf.is_contiguous() && buf.scalar_type() == at::kFloat);
  TORCH_CHECK(normsq_scratch.numel() >= 1 &&
              normsq_scratch.scalar_type() == at::kFloat);
  const long n = buf.numel();
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int grid = std::max(
      1L, std::min((long)kMaxGrid, (n + kBlock * 4 - 1) / (kBlock * 4)));
  CHECK_HIP(hipMemsetAsync(normsq_scratch.data_ptr(), 0, sizeof(float),
                           stream));
  hipLaunchKernelGGL(l2norm_sq_flat_kernel, dim3(grid), dim3(kBlock), 0,
                     stream, buf.data_ptr<float>(), n,
                     normsq_scratch.data_ptr<float>());
  hipLaunchKernelGGL(clip_scale_kernel, dim3(grid), dim3(kBlock), 0, stream,
                     buf.data_ptr<float>(), n,
                     normsq_scratch.data_ptr<float>(), (float)max_norm,
                     (float)eps);
  CHECK_HIP(hipGetLastError());
}

void scale_inplace(torch::Tensor buf, double scale) {
  TORCH_CHECK(buf.is_contiguous() && buf.scalar_type() == at::kFloat);
  const long n = buf.numel();
  const int grid =
      std::min((long)kMaxGrid, (n + kBlock * 4 - 1) / (kBlock * 4));
  hipLaunchKernelGGL(scale_inplace_kernel, dim3(std::max(grid, 1)),
                     dim3(kBlock), 0, c10::hip::getCurrentHIPStream().stream(),
                     buf.data_ptr<float>(), n, (float)scale);
  CHECK_HIP(hipGetLastError());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("build_sgd_table", &build_sgd_table,
        "Build device-side chunk table for fused SGD");
  m.def("multi_tensor_sgd", &multi_tensor_sgd, "Fused multi-tensor SGD");
  m.def("build_pack_table", &build_pack_table,
        "Build device-side chunk table for pack/unpack/norm");
  m.def("multi_tensor_pack", &multi_tensor_pack,
        "Gather grads into flat buffer (+cast/scale)");
  m.def("multi_tensor_unpack", &multi_tensor_unpack,
        "Scatter flat buffer into grads (+cast/scale)");
  m.def("l2norm_sq", &l2norm_sq, "Squared L2 norm over tensor list");
  m.def("scale_inplace", &scale_inplace, "In-place scale of a flat buffer");
  m.def("l2norm_clip_", &l2norm_clip_,
        "Device-side L2 norm clip of a flat buffer (no host sync)");
  m.attr("chunk_elems") = py::int_(kChunk);
}
