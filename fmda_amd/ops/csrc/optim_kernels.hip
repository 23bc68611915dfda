// Fused multi-tensor gradient-clip + Adam for MI355X (gfx950).
//
// Replaces the reference's clip_grad_norm_(params, clip) + Adam.step()
// (biGRU_model.py:208-210, notebook cell 29) with two launches over a
// packed chunk table:
//   1) global L2 norm^2 of all grads (block partials -> one atomicAdd)
//   2) Adam update with the clip scale min(1, clip/norm) folded into the
//      gradient read; m/v/param updated in one pass (fp32 masters).
// The chunk table lives in a single device buffer: per chunk
// {param*, grad*, m*, v*, numel_in_chunk} resolved host-side, so the
// kernels do no searching.
#include <hip/hip_runtime.h>

namespace fmda {

struct OptChunk {
    float* p;
    float* g;
    float* m;
    float* v;
    int n;
};

constexpr int OPT_NT = 256;  // chunking lives host-side (fmda_amd/optim.py)

__global__ void norm2_kernel(const OptChunk* __restrict__ chunks, int n_chunks,
                             float* __restrict__ out) {
    __shared__ float red[OPT_NT / 64];
    float s = 0.0f;
    for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
        const OptChunk ck = chunks[c];
        for (int i = threadIdx.x; i < ck.n; i += OPT_NT) {
            const float g = ck.g[i];
            s += g * g;
        }
    }
    s += __shfl_xor(s, 1);
    s += __shfl_xor(s, 2);
    s += __shfl_xor(s, 4);
    s += __shfl_xor(s, 8);
    s += __shfl_xor(s, 16);
    s += __shfl_xor(s, 32);
    const int wave = threadIdx.x >> 6;
    if ((threadIdx.x & 63) == 0) red[wave] = s;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t = 0.0f;
        for (int w = 0; w < OPT_NT / 64; ++w) t += red[w];
        atomicAdd(out, t);
    }
}

// scale = min(1, clip / (sqrt(norm2) + eps_n)); standard Adam with bias
// correction (torch.optim.Adam defaults).
__global__ void adam_kernel(const OptChunk* __restrict__ chunks, int n_chunks,
                            const float* __restrict__ norm2, float clip,
                            float lr, float beta1, float beta2, float eps,
                            float bc1, float bc2) {
    float scale = 1.0f;
    if (clip > 0.0f) {
        const float nrm = sqrtf(*norm2) + 1e-6f;
        if (nrm > clip) scale = clip / nrm;
    }
    for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
        const OptChunk ck = chunks[c];
        for (int i = threadIdx.x; i < ck.n; i += OPT_NT) {
            const float g = ck.g[i] * scale;
            const float m = beta1 * ck.m[i] + (1.0f - beta1) * g;
            const float v = beta2 * ck.v[i] + (1.0f - beta2) * g * g;
            ck.m[i] = m;
            ck.v[i] = v;
            const float mhat = m / bc1;
            const float vhat = v / bc2;
            ck.p[i] -= lr * mhat / (sqrtf(vhat) + eps);
        }
    }
}

extern "C" int fmda_opt_norm2_launch(const void* chunks, int n_chunks,
                                     float* out, hipStream_t stream) {
    const int grid = n_chunks < 2048 ? n_chunks : 2048;
    norm2_kernel<<<grid, OPT_NT, 0, stream>>>((const OptChunk*)chunks,
                                              n_chunks, out);
    return hipGetLastError() == hipSuccess ? 0 : -1;
}

extern "C" int fmda_opt_adam_launch(const void* chunks, int n_chunks,
                                    const float* norm2, float clip, float lr,
                                    float beta1, float beta2, float eps,
                                    float bc1, float bc2,
                                    hipStream_t stream) {
    const int grid = n_chunks < 2048 ? n_chunks : 2048;
    adam_kernel<<<grid, OPT_NT, 0, stream>>>((const OptChunk*)chunks, n_chunks,
                                             norm2, clip, lr, beta1, beta2,
                                             eps, bc1, bc2);
    return hipGetLastError() == hipSuccess ? 0 : -1;
}

}  // namespace fmda
