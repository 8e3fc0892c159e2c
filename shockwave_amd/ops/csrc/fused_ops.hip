// CDNA4 (gfx950 / MI355X) fused multi-tensor kernels for the
// dynamic-adaptation hot path of the shockwave_amd runtime.
//
// Named requirements (BASELINE.json north star; reference call sites in
// SURVEY.md §2.4):
//   * fused SGD-momentum / Adam optimizer step   (#4)
//   * Accordion grad-accumulate + per-tensor L2 norm (#5)
//   * GNS window-average + norm estimator        (#6)
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//   * wave = 64 lanes; 256-thread blocks (4 waves per CU workgroup)
//   * all kernels are HBM-bound: float4 (16 B/lane) vectorized main loops
//     (guide G13), grid built from 32K-element chunks so even an 11M-param
//     model yields ~340 workgroups (>256 CUs reachable, G11)
//   * reductions: 64-wide __shfl_down wave reduction -> LDS across the 4
//     waves -> ONE device-scope atomicAdd per block (G12)
//   * tensor/chunk metadata lives in a cached device buffer (no 4 KiB
//     kernel-arg limit, graph-capture safe: no allocation at launch time
//     once the cache is warm)
//
// Every kernel name is prefixed swq_ so rocprofv3 kernel traces attribute
// time unambiguously.

#include <hip/hip_runtime.h>

#define CHUNK_ELEMS 32768
#define BLOCK_THREADS 256
#define WAVE 64

// metadata layout in the int64 device buffer:
//   [0 .. NL*T)            : data pointers, list-major (addr[l*T + t])
//   [NL*T .. NL*T + T)     : numel per tensor
//   [NL*T + T .. +2*C)     : (tensor_idx, chunk_idx) per chunk
// where NL = number of tensor lists, T = tensor count, C = chunk count.

struct MetaView {
    const long long* buf;
    int num_tensors;
    int num_lists;
    __device__ inline void* addr(int list, int t) const {
        return (void*)buf[list * num_tensors + t];
    }
    __device__ inline long long numel(int t) const {
        return buf[num_lists * num_tensors + t];
    }
    __device__ inline int block_tensor(int b) const {
        return (int)buf[num_lists * num_tensors + num_tensors + 2 * b];
    }
    __device__ inline int block_chunk(int b) const {
        return (int)buf[num_lists * num_tensors + num_tensors + 2 * b + 1];
    }
};

// ---------------------------------------------------------------------------
// block-level sum reduction: wave shuffle then LDS
// ---------------------------------------------------------------------------
__device__ inline float block_reduce_sum(float v) {
    __shared__ float warp_sums[BLOCK_THREADS / WAVE];
    #pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        v += __shfl_down(v, off, WAVE);
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    if (lane == 0) warp_sums[wid] = v;
    __syncthreads();
    if (wid == 0) {
        v = (lane < BLOCK_THREADS / WAVE) ? warp_sums[lane] : 0.0f;
        #pragma unroll
        for (int off = (BLOCK_THREADS / WAVE) / 2; off > 0; off >>= 1)
            v += __shfl_down(v, off, WAVE);
    }
    return v;  // valid on thread 0
}

// ---------------------------------------------------------------------------
// fused SGD with momentum + weight decay (torch.optim.SGD semantics)
//   lists: 0=param, 1=grad, 2=momentum buffer
//   buf_initialized: 0 on the very first step (buf = d_p), 1 afterwards
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
swq_fused_sgd(const long long* meta, int num_tensors, float lr,
              float momentum, float dampening, float weight_decay,
              int nesterov, int buf_initialized) {
    MetaView mv{meta, num_tensors, 3};
    const int t = mv.block_tensor(blockIdx.x);
    const int c = mv.block_chunk(blockIdx.x);
    const long long base = (long long)c * CHUNK_ELEMS;
    const int n = (int)min((long long)CHUNK_ELEMS, mv.numel(t) - base);

    float* p = (float*)mv.addr(0, t) + base;
    const float* g = (const float*)mv.addr(1, t) + base;
    float* m = (float*)mv.addr(2, t) + base;

    const int vec_n = n / 4;
    float4* p4 = (float4*)p;
    const float4* g4 = (const float4*)g;
    float4* m4 = (float4*)m;

    for (int i = threadIdx.x; i < vec_n; i += BLOCK_THREADS) {
        float4 pv = p4[i];
        float4 gv = g4[i];
        float4 bv = m4[i];
        #pragma unroll
        for (int k = 0; k < 4; ++k) {
            float d_p = ((const float*)&gv)[k];
            float pk = ((const float*)&pv)[k];
            if (weight_decay != 0.0f) d_p += weight_decay * pk;
            float buf;
            if (momentum != 0.0f) {
                buf = buf_initialized
                          ? momentum * ((const float*)&bv)[k] + (1.0f - dampening) * d_p
                          : d_p;
                ((float*)&bv)[k] = buf;
                d_p = nesterov ? d_p + momentum * buf : buf;
            }
            ((float*)&pv)[k] = pk - lr * d_p;
        }
        p4[i] = pv;
        if (momentum != 0.0f) m4[i] = bv;
    }
    // scalar tail
    for (int i = vec_n * 4 + threadIdx.x; i < n; i += BLOCK_THREADS) {
        float d_p = g[i];
        float pk = p[i];
        if (weight_decay != 0.0f) d_p += weight_decay * pk;
        if (momentum != 0.0f) {
            float buf = buf_initialized
                            ? momentum * m[i] + (1.0f - dampening) * d_p
                            : d_p;
            m[i] = buf;
            d_p = nesterov ? d_p + momentum * buf : buf;
        }
        p[i] = pk - lr * d_p;
    }
}

// ---------------------------------------------------------------------------
// fused Adam / AdamW (torch.optim.Adam semantics, L2-style weight decay)
//   lists: 0=param, 1=grad, 2=exp_avg, 3=exp_avg_sq
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
swq_fused_adam(const long long* meta, int num_tensors, float lr, float beta1,
               float beta2, float eps, float weight_decay, int step,
               int adamw) {
    MetaView mv{meta, num_tensors, 4};
    const int t = mv.block_tensor(blockIdx.x);
    const int c = mv.block_chunk(blockIdx.x);
    const long long base = (long long)c * CHUNK_ELEMS;
    const int n = (int)min((long long)CHUNK_ELEMS, mv.numel(t) - base);

    float* p = (float*)mv.addr(0, t) + base;
    const float* g = (const float*)mv.addr(1, t) + base;
    float* m = (float*)mv.addr(2, t) + base;
    float* v = (float*)mv.addr(3, t) + base;

    const float bc1 = 1.0f - __powf(beta1, (float)step);
    const float bc2 = 1.0f - __powf(beta2, (float)step);
    const float step_size = lr / bc1;
    const float inv_sqrt_bc2 = rsqrtf(bc2);

    const int vec_n = n / 4;
    float4* p4 = (float4*)p;
    const float4* g4 = (const float4*)g;
    float4* m4 = (float4*)m;
    float4* v4 = (float4*)v;

    for (int i = threadIdx.x; i < vec_n; i += BLOCK_THREADS) {
        float4 pv = p4[i], gv = g4[i], mv_ = m4[i], vv = v4[i];
        #pragma unroll
        for (int k = 0; k < 4; ++k) {
            float pk = ((const float*)&pv)[k];
            float gk = ((const float*)&gv)[k];
            if (adamw) pk *= (1.0f - lr * weight_decay);
            else if (weight_decay != 0.0f) gk += weight_decay * pk;
            float mk = beta1 * ((const float*)&mv_)[k] + (1.0f - beta1) * gk;
            float vk = beta2 * ((const float*)&vv)[k] + (1.0f - beta2) * gk * gk;
            ((float*)&mv_)[k] = mk;
            ((float*)&vv)[k] = vk;
            float denom = sqrtf(vk) * inv_sqrt_bc2 + eps;
            ((float*)&pv)[k] = pk - step_size * mk / denom;
        }
        p4[i] = pv; m4[i] = mv_; v4[i] = vv;
    }
    for (int i = vec_n * 4 + threadIdx.x; i < n; i += BLOCK_THREADS) {
        float pk = p[i], gk = g[i];
        if (adamw) pk *= (1.0f - lr * weight_decay);
        else if (weight_decay != 0.0f) gk += weight_decay * pk;
        float mk = beta1 * m[i] + (1.0f - beta1) * gk;
        float vk = beta2 * v[i] + (1.0f - beta2) * gk * gk;
        m[i] = mk; v[i] = vk;
        p[i] = pk - step_size * mk / (sqrtf(vk) * inv_sqrt_bc2 + eps);
    }
}

// ---------------------------------------------------------------------------
// Accordion: multi-tensor accumulate  dst += src  (per-step grad accumulate)
//   lists: 0=dst(fp32 accumulator), 1=src(grad)
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
swq_multi_tensor_accum(const long long* meta, int num_tensors, float alpha) {
    MetaView mv{meta, num_tensors, 2};
    const int t = mv.block_tensor(blockIdx.x);
    const int c = mv.block_chunk(blockIdx.x);
    const long long base = (long long)c * CHUNK_ELEMS;
    const int n = (int)min((long long)CHUNK_ELEMS, mv.numel(t) - base);
    float* dst = (float*)mv.addr(0, t) + base;
    const float* src = (const float*)mv.addr(1, t) + base;

    const int vec_n = n / 4;
    float4* d4 = (float4*)dst;
    const float4* s4 = (const float4*)src;
    for (int i = threadIdx.x; i < vec_n; i += BLOCK_THREADS) {
        float4 d = d4[i], s = s4[i];
        d.x += alpha * s.x; d.y += alpha * s.y;
        d.z += alpha * s.z; d.w += alpha * s.w;
        d4[i] = d;
    }
    for (int i = vec_n * 4 + threadIdx.x; i < n; i += BLOCK_THREADS)
        dst[i] += alpha * src[i];
}

// ---------------------------------------------------------------------------
// Accordion: per-tensor squared L2 norms.  out[t] must be zeroed first.
//   lists: 0=src.  One atomicAdd per (block, tensor).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
swq_multi_tensor_l2norm_sq(const long long* meta, int num_tensors,
                           float* out) {
    MetaView mv{meta, num_tensors, 1};
    const int t = mv.block_tensor(blockIdx.x);
    const int c = mv.block_chunk(blockIdx.x);
    const long long base = (long long)c * CHUNK_ELEMS;
    const int n = (int)min((long long)CHUNK_ELEMS, mv.numel(t) - base);
    const float* src = (const float*)mv.addr(0, t) + base;

    float acc = 0.0f;
    const int vec_n = n / 4;
    const float4* s4 = (const float4*)src;
    for (int i = threadIdx.x; i < vec_n; i += BLOCK_THREADS) {
        float4 s = s4[i];
        acc += s.x * s.x + s.y * s.y + s.z * s.z + s.w * s.w;
    }
    for (int i = vec_n * 4 + threadIdx.x; i < n; i += BLOCK_THREADS)
        acc += src[i] * src[i];

    const float total = block_reduce_sum(acc);
    if (threadIdx.x == 0) atomicAdd(&out[t], total);
}

// ---------------------------------------------------------------------------
// GNS estimator: window-average gradient norm + current gradient norm.
//
//   grads: W pointers to flat fp32 gradients of length n (the sliding
//   window; grads[W-1] = the current gradient).
//   out[0] += || (1/W) sum_w g_w ||^2    (big-batch norm estimate)
//   out[1] += || g_{W-1} ||^2            (small-batch norm)
// Zero out[0:2] before launch.  Grid-stride, one pass over all W grads
// (W <= 8: per-lane accumulation in registers, 16 B loads per grad).
// ---------------------------------------------------------------------------
#define GNS_MAX_WINDOW 16
extern "C" __global__ void __launch_bounds__(BLOCK_THREADS)
swq_gns_window_stats(const long long* grad_ptrs, int window, long long n,
                     float* out) {
    float acc_big = 0.0f, acc_small = 0.0f;
    const float inv_w = 1.0f / (float)window;
    const long long vec_n = n / 4;
    const long long stride = (long long)gridDim.x * BLOCK_THREADS;
    const long long tid = (long long)blockIdx.x * BLOCK_THREADS + threadIdx.x;

    for (long long i = tid; i < vec_n; i += stride) {
        float4 mean = make_float4(0.f, 0.f, 0.f, 0.f);
        float4 last;
        for (int w = 0; w < window; ++w) {
            const float4 g = ((const float4*)grad_ptrs[w])[i];
            mean.x += g.x; mean.y += g.y; mean.z += g.z; mean.w += g.w;
            if (w == window - 1) last = g;
        }
        mean.x *= inv_w; mean.y *= inv_w; mean.z *= inv_w; mean.w *= inv_w;
        acc_big += mean.x * mean.x + mean.y * mean.y + mean.z * mean.z
                 + mean.w * mean.w;
        acc_small += last.x * last.x + last.y * last.y + last.z * last.z
                   + last.w * last.w;
    }
    for (long long i = vec_n * 4 + tid; i < n; i += stride) {
        float mean = 0.0f, last = 0.0f;
        for (int w = 0; w < window; ++w) {
            const float g = ((const float*)grad_ptrs[w])[i];
            mean += g;
            if (w == window - 1) last = g;
        }
        mean *= inv_w;
        acc_big += mean * mean;
        acc_small += last * last;
    }

    const float big = block_reduce_sum(acc_big);
    __syncthreads();  // reuse of the reduction LDS between the two calls
    const float small_ = block_reduce_sum(acc_small);
    if (threadIdx.x == 0) {
        atomicAdd(&out[0], big);
        atomicAdd(&out[1], small_);
    }
}

// ---------------------------------------------------------------------------
// host-side launchers (torch-independent; called from bindings.cpp)
// ---------------------------------------------------------------------------
extern "C" {

void swq_launch_fused_sgd(const long long* meta, int num_tensors,
                          int num_chunks, float lr, float momentum,
                          float dampening, float weight_decay, int nesterov,
                          int buf_initialized, void* stream) {
    hipLaunchKernelGGL(swq_fused_sgd, dim3(num_chunks), dim3(BLOCK_THREADS), 0,
                       (hipStream_t)stream, meta, num_tensors, lr, momentum,
                       dampening, weight_decay, nesterov, buf_initialized);
}

void swq_launch_fused_adam(const long long* meta, int num_tensors,
                           int num_chunks, float lr, float beta1, float beta2,
                           float eps, float weight_decay, int step, int adamw,
                           void* stream) {
    hipLaunchKernelGGL(swq_fused_adam, dim3(num_chunks), dim3(BLOCK_THREADS),
                       0, (hipStream_t)stream, meta, num_tensors, lr, beta1,
                       beta2, eps, weight_decay, step, adamw);
}

void swq_launch_multi_tensor_accum(const long long* meta, int num_tensors,
                                   int num_chunks, float alpha, void* stream) {
    hipLaunchKernelGGL(swq_multi_tensor_accum, dim3(num_chunks),
                       dim3(BLOCK_THREADS), 0, (hipStream_t)stream, meta,
                       num_tensors, alpha);
}

void swq_launch_multi_tensor_l2norm_sq(const long long* meta, int num_tensors,
                                       int num_chunks, float* out,
                                       void* stream) {
    hipLaunchKernelGGL(swq_multi_tensor_l2norm_sq, dim3(num_chunks),
                       dim3(BLOCK_THREADS), 0, (hipStream_t)stream, meta,
                       num_tensors, out);
}

void swq_launch_gns_window_stats(const long long* grad_ptrs, int window,
                                 long long n, float* out, void* stream) {
    long long vec_n = n / 4;
    int blocks = (int)((vec_n + BLOCK_THREADS - 1) / BLOCK_THREADS);
    if (blocks > 2048) blocks = 2048;  // grid-stride past this (guide G11)
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(swq_gns_window_stats, dim3(blocks),
                       dim3(BLOCK_THREADS), 0, (hipStream_t)stream, grad_ptrs,
                       window, n, out);
}

}  // extern "C"
