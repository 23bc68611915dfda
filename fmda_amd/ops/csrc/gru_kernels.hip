// Persistent biGRU recurrence kernels for MI355X (gfx950, CDNA4).
//
// Design (MI355X-first, not a port):
// - The time-batched input projections gi = x @ W_ih^T + b_ih for BOTH
//   directions are computed outside (one rocBLAS MFMA GEMM); these kernels
//   own the sequential part only: per timestep the recurrent GEMM
//   gh = h_{t-1} @ W_hh^T on MFMA (v_mfma_f32_16x16x32_bf16), the gate
//   sigmoid/tanh fusion, and the hidden-state update, with the recurrent
//   weights staged in LDS across all T timesteps (W in LDS for H <= 128;
//   L2-resident global reads beyond) and h kept in LDS in fp32 with a bf16
//   shadow for MFMA fragments.
// - One workgroup owns a tile of batch rows for one direction for the whole
//   sequence: no inter-workgroup communication, no grid sync. Both
//   directions launch in one grid (blockIdx.y = direction).
// - The backward (BPTT) kernel recomputes the gates from gi + a fresh
//   recurrent GEMM (cheaper than materializing activations: the op is
//   HBM-bound), producing dGi (input-projection grads, consumed by torch
//   autograd for dW_ih/db_ih/dx) and dGh (recurrent-gate grads, reduced to
//   dW_hh/db_hh by one rocBLAS GEMM outside).
// - fp32 instantiations use a plain VALU dot-product path with the same
//   phase structure and C-tile ownership; they are the on-GPU numerics
//   oracle, validated against the PyTorch fp32 reference.
//
// Gate math follows the PyTorch packed (r, z, n) convention
// (reference biGRU_model.py:54-56 uses nn.GRU):
//   r = sigmoid(i_r + W_hr h + b_hr)
//   z = sigmoid(i_z + W_hz h + b_hz)
//   n = tanh(i_n + r * (W_hn h + b_hn))
//   h' = (1 - z) * n + z * h
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define FMDA_DEV __device__ __forceinline__

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));
typedef short short8_t __attribute__((ext_vector_type(8)));

namespace fmda {

FMDA_DEV float sigmoidf(float x) { return 1.0f / (1.0f + expf(-x)); }

template <typename T> FMDA_DEV float to_f32(T v);
template <> FMDA_DEV float to_f32<float>(float v) { return v; }
template <> FMDA_DEV float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
    return __bfloat162float(v);
}
template <typename T> FMDA_DEV T from_f32(float v);
template <> FMDA_DEV float from_f32<float>(float v) { return v; }
template <> FMDA_DEV __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
    return __float2bfloat16(v);
}

// ---------------------------------------------------------------------------
// Cooperative tile copy helpers (256 threads), vectorized by 16-byte chunks
// (row byte counts are multiples of 16 for every supported Hp). Global rows
// may be masked by `rows_valid`: out-of-range rows are skipped — their LDS
// content is stale/garbage but never observed in the final output (batch
// rows are independent in a GRU recurrence).
// ---------------------------------------------------------------------------

struct alignas(16) chunk16 { unsigned int u[4]; };

// LDS tile [rows][cols] <- global rows at row_stride (elements).
template <typename T, int ROWS>
FMDA_DEV void stage_tile(T* __restrict__ lds, const T* __restrict__ gp,
                         int cols, int lds_pitch, long row_stride,
                         int rows_valid, int tid) {
    const int cpr = (cols * (int)sizeof(T)) / 16;  // chunks per row
    const int total = ROWS * cpr;
    for (int c = tid; c < total; c += 256) {
        const int r = c / cpr;
        const int jc = c % cpr;
        if (r < rows_valid) {
            ((chunk16*)((char*)lds + (long)r * lds_pitch * sizeof(T)))[jc] =
                ((const chunk16*)((const char*)gp +
                                  (long)r * row_stride * sizeof(T)))[jc];
        }
    }
}

template <typename T, int ROWS>
FMDA_DEV void store_tile(const T* __restrict__ lds, T* __restrict__ gp,
                         int cols, int lds_pitch, long row_stride,
                         int rows_valid, int tid) {
    const int cpr = (cols * (int)sizeof(T)) / 16;
    const int total = ROWS * cpr;
    for (int c = tid; c < total; c += 256) {
        const int r = c / cpr;
        const int jc = c % cpr;
        if (r < rows_valid) {
            ((chunk16*)((char*)gp + (long)r * row_stride * sizeof(T)))[jc] =
                ((const chunk16*)((const char*)lds +
                                  (long)r * lds_pitch * sizeof(T)))[jc];
        }
    }
}

template <typename T, int ROWS>
FMDA_DEV void zero_tile(T* __restrict__ lds, int cols, int lds_pitch, int tid) {
    for (int c = tid; c < ROWS * lds_pitch; c += 256) lds[c] = from_f32<T>(0.0f);
    (void)cols;
}

// fp32 LDS tile += bf16/f32 global tile (used for dh += dOut[t]).
template <typename T, int ROWS>
FMDA_DEV void accum_tile_f32(float* __restrict__ lds, const T* __restrict__ gp,
                             int cols, int lds_pitch, long row_stride,
                             int rows_valid, int tid) {
    const int total = ROWS * cols;
    for (int c = tid; c < total; c += 256) {
        const int r = c / cols;
        const int j = c % cols;
        if (r < rows_valid)
            lds[r * lds_pitch + j] += to_f32<T>(gp[(long)r * row_stride + j]);
    }
}

// ---------------------------------------------------------------------------
// MFMA / VALU recurrent GEMM phase.
//
// C-tile ownership (v_mfma_f32_16x16x32_bf16 layout; ck_tile
// warp_gemm_attribute_mfma_impl.hpp M16N16K32 constants):
//   A (M=16, K=32): lane l holds A[l%16][8*(l/16) + e], e = 0..7 (contiguous)
//   B (K=32, N=16): lane l holds B[8*(l/16) + e][l%16]
//   C (16x16):      lane l, reg v holds C[4*(l/16) + v][l%16]
// The fp32 path computes the same owned C elements with plain dots so the
// fused gate phase is identical for both dtypes.
// ---------------------------------------------------------------------------

// Per-wave accumulate of gh tiles for column-tile `ct` of every gate.
// hb: LDS [BT][Hp] fragment source (bf16 shadow of h, or fp32 h directly).
// w_row(n) returns pointer to row n (length Hp) of W_hh (LDS or global).
template <int BT, int Hp>
FMDA_DEV void gemm_ct_bf16(const __hip_bfloat16* __restrict__ hb, int hb_pitch,
                           const __hip_bfloat16* __restrict__ wbase,
                           long w_row_stride, int ct, int lane,
                           f32x4_t acc[3][BT / 16]) {
    constexpr int MT = BT / 16;
    const int arow = lane & 15;
    const int koff = 8 * (lane >> 4);
    const int jcol = ct * 16 + (lane & 15);
#pragma unroll
    for (int kk = 0; kk < Hp / 32; ++kk) {
        const int kbase = 32 * kk + koff;
        bf16x8_t a[MT];
#pragma unroll
        for (int m = 0; m < MT; ++m)
            a[m] = *(const bf16x8_t*)&hb[(16 * m + arow) * hb_pitch + kbase];
#pragma unroll
        for (int g = 0; g < 3; ++g) {
            const long n = g * Hp + jcol;
            bf16x8_t b = *(const bf16x8_t*)&wbase[n * w_row_stride + kbase];
#pragma unroll
            for (int m = 0; m < MT; ++m)
                acc[g][m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a[m], b, acc[g][m], 0, 0, 0);
        }
    }
}

template <int BT, int Hp>
FMDA_DEV void gemm_ct_f32(const float* __restrict__ hf, int hb_pitch,
                          const float* __restrict__ wbase,
                          long w_row_stride, int ct, int lane,
                          f32x4_t acc[3][BT / 16]) {
    constexpr int MT = BT / 16;
    const int jcol = ct * 16 + (lane & 15);
#pragma unroll
    for (int g = 0; g < 3; ++g) {
        const float* wrow = wbase + (long)(g * Hp + jcol) * w_row_stride;
#pragma unroll
        for (int m = 0; m < MT; ++m) {
#pragma unroll
            for (int e = 0; e < 4; ++e) {
                const int row = 16 * m + 4 * (lane >> 4) + e;
                float s = acc[g][m][e];
                for (int k = 0; k < Hp; ++k)
                    s += hf[row * hb_pitch + k] * wrow[k];
                acc[g][m][e] = s;
            }
        }
    }
}

// dh GEMM: dh[b][j] += sum_n dgh[b][n] * W[n][j]  (K = 3*Hp, N = Hp).
// A = dgh (LDS, row-major, contiguous k). B[k=n][j] = W[n][j]: strided reads.
template <int BT, int Hp>
FMDA_DEV void gemm_dh_bf16(const __hip_bfloat16* __restrict__ dgh, int dgh_pitch,
                           const __hip_bfloat16* __restrict__ wbase,
                           long w_row_stride, int ct, int lane,
                           f32x4_t acc[BT / 16]) {
    constexpr int MT = BT / 16;
    const int arow = lane & 15;
    const int koff = 8 * (lane >> 4);
    const int jcol = ct * 16 + (lane & 15);
#pragma unroll
    for (int kk = 0; kk < (3 * Hp) / 32; ++kk) {
        const int kbase = 32 * kk + koff;
        bf16x8_t a[MT];
#pragma unroll
        for (int m = 0; m < MT; ++m)
            a[m] = *(const bf16x8_t*)&dgh[(16 * m + arow) * dgh_pitch + kbase];
        bf16x8_t b;
        const __bf16* wb = (const __bf16*)wbase;
#pragma unroll
        for (int e = 0; e < 8; ++e)
            b[e] = wb[(long)(kbase + e) * w_row_stride + jcol];
#pragma unroll
        for (int m = 0; m < MT; ++m)
            acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[m], b, acc[m],
                                                             0, 0, 0);
    }
}

template <int BT, int Hp>
FMDA_DEV void gemm_dh_f32(const float* __restrict__ dgh, int dgh_pitch,
                          const float* __restrict__ wbase,
                          long w_row_stride, int ct, int lane,
                          f32x4_t acc[BT / 16]) {
    constexpr int MT = BT / 16;
    const int jcol = ct * 16 + (lane & 15);
#pragma unroll
    for (int m = 0; m < MT; ++m) {
#pragma unroll
        for (int e = 0; e < 4; ++e) {
            const int row = 16 * m + 4 * (lane >> 4) + e;
            float s = acc[m][e];
            for (int n = 0; n < 3 * Hp; ++n)
                s += dgh[row * dgh_pitch + n] * wbase[(long)n * w_row_stride + jcol];
            acc[m][e] = s;
        }
    }
}

// ---------------------------------------------------------------------------
// Forward kernel.
//
// gi:    (B, T, n_dir*3Hp)  input projections incl. b_ih (dtype T)
// w:     (n_dir, 3Hp, Hp)   recurrent weights, row-major [n][k] (dtype T)
// bhh:   (n_dir, 3Hp)       recurrent bias (fp32)
// out:   (B, T, n_dir*Hp)   hidden states (dtype T; direction-concat layout)
// hlast: (n_dir, B, Hp)     final hidden state (fp32)
// grid:  (ceil(B/BT), n_dir); block: 256 threads; direction 1 runs reversed.
// ---------------------------------------------------------------------------
template <typename T, int BT, int Hp, bool WLDS>
__global__ __launch_bounds__(256) void gru_fwd_kernel(
    const T* __restrict__ gi, const T* __restrict__ w,
    const float* __restrict__ bhh, T* __restrict__ out,
    float* __restrict__ hlast, int B, int Tseq, int n_dir) {
    constexpr int MT = BT / 16;
    constexpr int NCT = Hp / 16;          // column tiles per gate
    constexpr int CPW = (NCT + 3) / 4;    // column tiles per wave (max)
    constexpr bool IS_BF16 = !__is_same(T, float);
    // Padded LDS row pitches (+16 B per row): the fragment reads hit 16
    // distinct rows at one 16-B column slot; an even power-of-two row pitch
    // puts the whole lane group on one LDS bank slot (16-way conflict), the
    // odd-slot pitch spreads it over all 16 slots (conflict-free).
    constexpr int PADE = 16 / (int)sizeof(T);
    constexpr int WP = Hp + PADE;           // pitch of Hp-wide T rows
    constexpr int GP3 = 3 * Hp + PADE;      // pitch of 3Hp-wide T rows
    constexpr int HFP = Hp + 4;             // pitch of Hp-wide fp32 rows

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int b0 = blockIdx.x * BT;
    const int dir = blockIdx.y;
    const bool rev = (dir == 1);
    const int rows_valid = min(BT, B - b0);

    extern __shared__ __attribute__((aligned(16))) char smem[];
    char* p = smem;
    T* w_s = nullptr;
    if (WLDS) { w_s = (T*)p; p += sizeof(T) * 3 * Hp * WP; }
    float* hf_s = (float*)p; p += sizeof(float) * BT * HFP;
    __hip_bfloat16* hb_s = nullptr;
    if (IS_BF16) { hb_s = (__hip_bfloat16*)p; p += 2 * BT * WP; }
    T* gi_s = (T*)p; p += sizeof(T) * BT * GP3;
    float* bhh_s = (float*)p;

    const T* wdir = w + (long)dir * 3 * Hp * Hp;
    if (WLDS)
        stage_tile<T, 3 * Hp>(w_s, wdir, Hp, WP, Hp, 3 * Hp, tid);
    for (int c = tid; c < 3 * Hp; c += 256)
        bhh_s[c] = bhh[(long)dir * 3 * Hp + c];
    for (int c = tid; c < BT * HFP; c += 256) hf_s[c] = 0.0f;
    if (IS_BF16)
        for (int c = tid; c < BT * WP; c += 256)
            hb_s[c] = __float2bfloat16(0.0f);

    const long gi_row = (long)Tseq * n_dir * 3 * Hp;   // per batch row
    const long out_row = (long)Tseq * n_dir * Hp;
    const T* gi_b = gi + (long)b0 * gi_row + (long)dir * 3 * Hp;
    T* out_b = out + (long)b0 * out_row + (long)dir * Hp;

    // stage gi for step 0
    {
        const int tt = rev ? (Tseq - 1) : 0;
        stage_tile<T, BT>(gi_s, gi_b + (long)tt * n_dir * 3 * Hp, 3 * Hp, GP3,
                          gi_row, rows_valid, tid);
    }
    __syncthreads();

    const T* wfrag = WLDS ? w_s : wdir;
    const long wstride = WLDS ? WP : Hp;  // row stride in elements

    for (int u = 0; u < Tseq; ++u) {
        const int tt = rev ? (Tseq - 1 - u) : u;
        // ---- phase A: recurrent GEMM gh = h @ W^T (per-wave column tiles)
        f32x4_t acc[CPW][3][MT];
#pragma unroll
        for (int i = 0; i < CPW; ++i)
#pragma unroll
            for (int g = 0; g < 3; ++g)
#pragma unroll
                for (int m = 0; m < MT; ++m) acc[i][g][m] = f32x4_t{0.f};
#pragma unroll
        for (int i = 0; i < CPW; ++i) {
            const int ct = wave + 4 * i;
            if (ct < NCT) {
                if constexpr (IS_BF16)
                    gemm_ct_bf16<BT, Hp>(hb_s, WP, wfrag, wstride, ct, lane,
                                         acc[i]);
                else
                    gemm_ct_f32<BT, Hp>(hf_s, HFP, wfrag, wstride, ct, lane,
                                        acc[i]);
            }
        }
        __syncthreads();  // all GEMM reads of h done

        // ---- phase B: fused gates + h update (owned (b, j) elements)
#pragma unroll
        for (int i = 0; i < CPW; ++i) {
            const int ct = wave + 4 * i;
            if (ct >= NCT) continue;
            const int j = ct * 16 + (lane & 15);
#pragma unroll
            for (int m = 0; m < MT; ++m) {
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    const int b = 16 * m + 4 * (lane >> 4) + e;
                    const float gr = acc[i][0][m][e] + bhh_s[j];
                    const float gz = acc[i][1][m][e] + bhh_s[Hp + j];
                    const float hn = acc[i][2][m][e] + bhh_s[2 * Hp + j];
                    const float ir = to_f32<T>(gi_s[b * GP3 + j]);
                    const float iz = to_f32<T>(gi_s[b * GP3 + Hp + j]);
                    const float in_ = to_f32<T>(gi_s[b * GP3 + 2 * Hp + j]);
                    const float r = sigmoidf(ir + gr);
                    const float z = sigmoidf(iz + gz);
                    const float n = tanhf(in_ + r * hn);
                    const float hprev = hf_s[b * HFP + j];
                    const float hnew = (1.0f - z) * n + z * hprev;
                    hf_s[b * HFP + j] = hnew;
                    if constexpr (IS_BF16)
                        hb_s[b * WP + j] = __float2bfloat16(hnew);
                }
            }
        }
        __syncthreads();  // h_t complete; gi_s free

        // ---- phase C: write out[t]; stage gi for t+1
        if constexpr (IS_BF16) {
            store_tile<__hip_bfloat16, BT>(
                hb_s, (__hip_bfloat16*)(out_b + (long)tt * n_dir * Hp), Hp, WP,
                out_row, rows_valid, tid);
        } else {
            store_tile<float, BT>(hf_s, (float*)(out_b + (long)tt * n_dir * Hp),
                                  Hp, HFP, out_row, rows_valid, tid);
        }
        if (u + 1 < Tseq) {
            const int ttn = rev ? (Tseq - 2 - u) : (u + 1);
            stage_tile<T, BT>(gi_s, gi_b + (long)ttn * n_dir * 3 * Hp, 3 * Hp,
                              GP3, gi_row, rows_valid, tid);
        }
        __syncthreads();
    }

    // final hidden state (fp32)
    float* hl = hlast + ((long)dir * B + b0) * Hp;
    for (int c = tid; c < BT * Hp; c += 256) {
        const int r = c / Hp;
        if (r < rows_valid) hl[(long)r * Hp + (c % Hp)] = hf_s[r * HFP + (c % Hp)];
    }
}

// ---------------------------------------------------------------------------
// Backward (BPTT) kernel.
//
// Recomputes gates from gi + fresh recurrent GEMM on the stored h (out),
// produces dGi and dGh, and carries dh in fp32 LDS across timesteps.
// dW_hh/db_hh are reduced outside from dGh (plain GEMM).
// ---------------------------------------------------------------------------
template <typename T, int BT, int Hp, bool WLDS>
__global__ __launch_bounds__(256) void gru_bwd_kernel(
    const T* __restrict__ gi, const T* __restrict__ w,
    const float* __restrict__ bhh, const T* __restrict__ out,
    const T* __restrict__ dout, const float* __restrict__ dhT,
    T* __restrict__ dgi, T* __restrict__ dgh, float* __restrict__ dh0,
    float* __restrict__ dbhh, int B, int Tseq, int n_dir) {
    constexpr int MT = BT / 16;
    constexpr int NCT = Hp / 16;
    constexpr int CPW = (NCT + 3) / 4;
    constexpr bool IS_BF16 = !__is_same(T, float);
    constexpr int PADE = 16 / (int)sizeof(T);
    constexpr int WP = Hp + PADE;
    constexpr int GP3 = 3 * Hp + PADE;
    constexpr int HFP = Hp + 4;

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int b0 = blockIdx.x * BT;
    const int dir = blockIdx.y;
    const bool rev = (dir == 1);
    const int rows_valid = min(BT, B - b0);

    extern __shared__ __attribute__((aligned(16))) char smem[];
    char* p = smem;
    T* w_s = nullptr;
    if (WLDS) { w_s = (T*)p; p += sizeof(T) * 3 * Hp * WP; }
    float* dh_s = (float*)p; p += sizeof(float) * BT * HFP;
    T* hb_s = (T*)p; p += sizeof(T) * BT * WP;      // h_{t-1} (from out)
    T* gi_s = (T*)p; p += sizeof(T) * BT * GP3;  // gi[t], then overwritten by dGi
    T* dgh_s = (T*)p; p += sizeof(T) * BT * GP3;
    float* bhh_s = (float*)p;

    const T* wdir = w + (long)dir * 3 * Hp * Hp;
    if (WLDS)
        stage_tile<T, 3 * Hp>(w_s, wdir, Hp, WP, Hp, 3 * Hp, tid);
    for (int c = tid; c < 3 * Hp; c += 256)
        bhh_s[c] = bhh[(long)dir * 3 * Hp + c];

    const long gi_row = (long)Tseq * n_dir * 3 * Hp;
    const long out_row = (long)Tseq * n_dir * Hp;
    const T* gi_b = gi + (long)b0 * gi_row + (long)dir * 3 * Hp;
    T* dgi_b = dgi + (long)b0 * gi_row + (long)dir * 3 * Hp;
    T* dgh_b = dgh + (long)b0 * gi_row + (long)dir * 3 * Hp;
    const T* out_b = out + (long)b0 * out_row + (long)dir * Hp;
    const T* dout_b = dout + (long)b0 * out_row + (long)dir * Hp;

    // dh carry init: dh = dhT; then += dOut[t] as each step is staged.
    {
        const float* hT = dhT + ((long)dir * B + b0) * Hp;
        for (int c = tid; c < BT * Hp; c += 256) {
            const int r = c / Hp;
            dh_s[r * HFP + (c % Hp)] =
                (r < rows_valid) ? hT[(long)r * Hp + (c % Hp)] : 0.0f;
        }
    }

    // Stage step u = Tseq-1: gi[tt], h_prev = out[tt_prev] or 0, dh += dout[tt]
    {
        const int u = Tseq - 1;
        const int tt = rev ? (Tseq - 1 - u) : u;
        stage_tile<T, BT>(gi_s, gi_b + (long)tt * n_dir * 3 * Hp, 3 * Hp, GP3,
                          gi_row, rows_valid, tid);
        if (u > 0) {
            const int ttp = rev ? (Tseq - u) : (u - 1);
            stage_tile<T, BT>(hb_s, out_b + (long)ttp * n_dir * Hp, Hp, WP,
                              out_row, rows_valid, tid);
        } else {
            zero_tile<T, BT>(hb_s, Hp, WP, tid);
        }
        accum_tile_f32<T, BT>(dh_s, dout_b + (long)tt * n_dir * Hp, Hp, HFP,
                              out_row, rows_valid, tid);
    }
    __syncthreads();

    const T* wfrag = WLDS ? w_s : wdir;
    const long wstride = WLDS ? WP : Hp;
    // per-lane db_hh partials accumulated across all T (summed over b rows)
    float dbacc[CPW][3];
#pragma unroll
    for (int i = 0; i < CPW; ++i)
#pragma unroll
        for (int g = 0; g < 3; ++g) dbacc[i][g] = 0.0f;

    for (int u = Tseq - 1; u >= 0; --u) {
        const int tt = rev ? (Tseq - 1 - u) : u;

        // ---- phase A: recompute gh, fuse gate grads (per owned column tile)
#pragma unroll
        for (int i = 0; i < CPW; ++i) {
            const int ct = wave + 4 * i;
            if (ct >= NCT) continue;
            f32x4_t acc[3][MT];
#pragma unroll
            for (int g = 0; g < 3; ++g)
#pragma unroll
                for (int m = 0; m < MT; ++m) acc[g][m] = f32x4_t{0.f};
            if constexpr (IS_BF16)
                gemm_ct_bf16<BT, Hp>(hb_s, WP, wfrag, wstride, ct, lane, acc);
            else
                gemm_ct_f32<BT, Hp>((const float*)hb_s, WP, wfrag, wstride, ct,
                                    lane, acc);
            const int j = ct * 16 + (lane & 15);
#pragma unroll
            for (int m = 0; m < MT; ++m) {
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    const int b = 16 * m + 4 * (lane >> 4) + e;
                    const float gr = acc[0][m][e] + bhh_s[j];
                    const float gz = acc[1][m][e] + bhh_s[Hp + j];
                    const float hn = acc[2][m][e] + bhh_s[2 * Hp + j];
                    const float ir = to_f32<T>(gi_s[b * GP3 + j]);
                    const float iz = to_f32<T>(gi_s[b * GP3 + Hp + j]);
                    const float in_ = to_f32<T>(gi_s[b * GP3 + 2 * Hp + j]);
                    const float r = sigmoidf(ir + gr);
                    const float z = sigmoidf(iz + gz);
                    const float n = tanhf(in_ + r * hn);
                    const float hprev = to_f32<T>(hb_s[b * WP + j]);
                    const float dht = dh_s[b * HFP + j];
                    const bool live = (b < rows_valid);
                    const float dz_pre = live ? dht * (hprev - n) * z * (1.0f - z) : 0.0f;
                    const float dn_pre = live ? dht * (1.0f - z) * (1.0f - n * n) : 0.0f;
                    const float dr_pre = dn_pre * hn * r * (1.0f - r);
                    const float dhn = dn_pre * r;
                    // dGi (packed r,z,n pre-activation grads) -> reuse gi_s
                    gi_s[b * GP3 + j] = from_f32<T>(dr_pre);
                    gi_s[b * GP3 + Hp + j] = from_f32<T>(dz_pre);
                    gi_s[b * GP3 + 2 * Hp + j] = from_f32<T>(dn_pre);
                    // dGh differs in the n block only
                    dgh_s[b * GP3 + j] = from_f32<T>(dr_pre);
                    dgh_s[b * GP3 + Hp + j] = from_f32<T>(dz_pre);
                    dgh_s[b * GP3 + 2 * Hp + j] = from_f32<T>(dhn);
                    // direct part of dh_{t-1}; GEMM part added in phase B
                    dh_s[b * HFP + j] = dht * z;
                    // db_hh partials (summed in fp32 across b and t)
                    dbacc[i][0] += dr_pre;
                    dbacc[i][1] += dz_pre;
                    dbacc[i][2] += dhn;
                }
            }
        }
        __syncthreads();  // dgh_s complete

        // ---- phase B: dh_{t-1} += dGh @ W
#pragma unroll
        for (int i = 0; i < CPW; ++i) {
            const int ct = wave + 4 * i;
            if (ct >= NCT) continue;
            f32x4_t acc2[MT];
#pragma unroll
            for (int m = 0; m < MT; ++m) acc2[m] = f32x4_t{0.f};
            if constexpr (IS_BF16)
                gemm_dh_bf16<BT, Hp>((const __hip_bfloat16*)dgh_s, GP3, wfrag,
                                     wstride, ct, lane, acc2);
            else
                gemm_dh_f32<BT, Hp>((const float*)dgh_s, GP3, wfrag, wstride,
                                    ct, lane, acc2);
            const int j = ct * 16 + (lane & 15);
#pragma unroll
            for (int m = 0; m < MT; ++m)
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    const int b = 16 * m + 4 * (lane >> 4) + e;
                    dh_s[b * HFP + j] += acc2[m][e];
                }
        }
        __syncthreads();

        // ---- phase C: write dGi/dGh; stage next step
        store_tile<T, BT>(gi_s, dgi_b + (long)tt * n_dir * 3 * Hp, 3 * Hp, GP3,
                          gi_row, rows_valid, tid);
        store_tile<T, BT>(dgh_s, dgh_b + (long)tt * n_dir * 3 * Hp, 3 * Hp,
                          GP3, gi_row, rows_valid, tid);
        if (u > 0) {
            const int un = u - 1;
            const int ttn = rev ? (Tseq - 1 - un) : un;
            stage_tile<T, BT>(gi_s, gi_b + (long)ttn * n_dir * 3 * Hp, 3 * Hp,
                              GP3, gi_row, rows_valid, tid);
            if (un > 0) {
                const int ttp = rev ? (Tseq - un) : (un - 1);
                stage_tile<T, BT>(hb_s, out_b + (long)ttp * n_dir * Hp, Hp, WP,
                                  out_row, rows_valid, tid);
            } else {
                zero_tile<T, BT>(hb_s, Hp, WP, tid);
            }
            accum_tile_f32<T, BT>(dh_s, dout_b + (long)ttn * n_dir * Hp, Hp,
                                  HFP, out_row, rows_valid, tid);
        }
        __syncthreads();
    }

    // dh0 (fp32)
    float* d0 = dh0 + ((long)dir * B + b0) * Hp;
    for (int c = tid; c < BT * Hp; c += 256) {
        const int r = c / Hp;
        if (r < rows_valid) d0[(long)r * Hp + (c % Hp)] = dh_s[r * HFP + (c % Hp)];
    }

    // db_hh: reduce the 4 row-quarter lanes (same j = lane&15), then one
    // atomicAdd per (gate, j) per block.
#pragma unroll
    for (int i = 0; i < CPW; ++i) {
        const int ct = wave + 4 * i;
        if (ct >= NCT) continue;
#pragma unroll
        for (int g = 0; g < 3; ++g) {
            float v = dbacc[i][g];
            v += __shfl_xor(v, 16);
            v += __shfl_xor(v, 32);
            if ((lane >> 4) == 0)
                atomicAdd(&dbhh[(long)dir * 3 * Hp + g * Hp + ct * 16 +
                                (lane & 15)], v);
        }
    }
}

// ---------------------------------------------------------------------------
// MFMA layout self-test: C (16x16) = A (16x32) @ B (32x16) in bf16 with the
// fragment layout assumed above. A GPU test compares it against torch matmul
// so a layout regression fails loudly instead of silently transposing.
// ---------------------------------------------------------------------------
__global__ void mfma_selftest_kernel(const __hip_bfloat16* __restrict__ A,
                                     const __hip_bfloat16* __restrict__ Bm,
                                     float* __restrict__ C) {
    const int lane = threadIdx.x & 63;
    bf16x8_t a, b;
    const __bf16* ap = (const __bf16*)A;
    const __bf16* bp = (const __bf16*)Bm;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        a[e] = ap[(lane & 15) * 32 + 8 * (lane >> 4) + e];   // A[i][k] row-major
        b[e] = bp[(8 * (lane >> 4) + e) * 16 + (lane & 15)]; // B[k][j] row-major
    }
    f32x4_t acc = f32x4_t{0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
    for (int v = 0; v < 4; ++v)
        C[(4 * (lane >> 4) + v) * 16 + (lane & 15)] = acc[v];
}

}  // namespace fmda

// ---------------------------------------------------------------------------
// Host launchers (exported; bound to torch in bindings.cpp).
// ---------------------------------------------------------------------------

namespace fmda {

struct LaunchCfg { int bt; bool wlds; };

// Tile/LDS policy per (dtype, Hp): chosen so total LDS stays under 160 KiB
// with the recurrent weights resident whenever they fit.
static inline LaunchCfg fwd_cfg(bool bf16, int Hp) {
    if (bf16) {
        if (Hp <= 128) return {32, true};
        if (Hp == 256) return {32, false};
        return {16, false};
    }
    if (Hp <= 128) return {32, false};
    return {16, false};
}
static inline LaunchCfg bwd_cfg(bool bf16, int Hp) {
    if (bf16) {
        if (Hp <= 64) return {32, true};
        if (Hp == 128) return {16, true};
        if (Hp == 256) return {32, false};
        return {16, false};
    }
    if (Hp <= 128) return {32, false};
    return {16, false};
}

// NOTE: must mirror the in-kernel padded pitches (WP/GP3/HFP).
static inline size_t fwd_lds_bytes(bool bf16, int Hp, int bt, bool wlds) {
    const size_t es = bf16 ? 2 : 4;
    const size_t pade = 16 / es;
    const size_t wp = Hp + pade, gp3 = 3 * Hp + pade, hfp = Hp + 4;
    size_t s = 0;
    if (wlds) s += es * 3 * Hp * wp;
    s += 4 * (size_t)bt * hfp;             // hf_s
    if (bf16) s += 2 * (size_t)bt * wp;    // hb_s
    s += es * (size_t)bt * gp3;            // gi_s
    s += 4 * (size_t)3 * Hp;               // bhh_s
    return s;
}
static inline size_t bwd_lds_bytes(bool bf16, int Hp, int bt, bool wlds) {
    const size_t es = bf16 ? 2 : 4;
    const size_t pade = 16 / es;
    const size_t wp = Hp + pade, gp3 = 3 * Hp + pade, hfp = Hp + 4;
    size_t s = 0;
    if (wlds) s += es * 3 * Hp * wp;
    s += 4 * (size_t)bt * hfp;             // dh_s
    s += es * (size_t)bt * wp;             // hb_s
    s += 2 * es * (size_t)bt * gp3;        // gi_s + dgh_s
    s += 4 * (size_t)3 * Hp;               // bhh_s
    return s;
}

#define FMDA_DISPATCH_HP(HP_VAL, FN)                                          \
    switch (HP_VAL) {                                                         \
        case 16: FN(16); break;                                               \
        case 32: FN(32); break;                                               \
        case 64: FN(64); break;                                               \
        case 128: FN(128); break;                                             \
        case 256: FN(256); break;                                             \
        case 512: FN(512); break;                                             \
        default: return -1;                                                   \
    }

extern "C" int fmda_gru_fwd_launch(int is_bf16, int Hp, const void* gi,
                                   const void* w, const float* bhh, void* out,
                                   float* hlast, int B, int Tseq, int n_dir,
                                   hipStream_t stream) {
    const LaunchCfg cfg = fwd_cfg(is_bf16, Hp);
    const size_t lds = fwd_lds_bytes(is_bf16, Hp, cfg.bt, cfg.wlds);
    if (lds > 160 * 1024) return -2;
    const dim3 grid((B + cfg.bt - 1) / cfg.bt, n_dir);
    hipError_t err = hipSuccess;

#define FWD_CASE(HPV)                                                          \
    do {                                                                       \
        if (is_bf16) {                                                         \
            if (cfg.bt == 32 && cfg.wlds) {                                    \
                auto k = gru_fwd_kernel<__hip_bfloat16, 32, HPV, true>;        \
                hipFuncSetAttribute((const void*)k,                            \
                    hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);     \
                k<<<grid, 256, lds, stream>>>((const __hip_bfloat16*)gi,       \
                    (const __hip_bfloat16*)w, bhh, (__hip_bfloat16*)out,       \
                    hlast, B, Tseq, n_dir);                                    \
            } else if (cfg.bt == 32) {                                         \
                auto k = gru_fwd_kernel<__hip_bfloat16, 32, HPV, false>;       \
                hipFuncSetAttribute((const void*)k,                            \
                    hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);     \
                k<<<grid, 256, lds, stream>>>((const __hip_bfloat16*)gi,       \
                    (const __hip_bfloat16*)w, bhh, (__hip_bfloat16*)out,       \
                    hlast, B, Tseq, n_dir);                                    \
            } else {                                                           \
                auto k = gru_fwd_kernel<__hip_bfloat16, 16, HPV, false>;       \
                hipFuncSetAttribute((const void*)k,                            \
                    hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);     \
                k<<<grid, 256, lds, stream>>>((const __hip_bfloat16*)gi,       \
                    (const __hip_bfloat16*)w, bhh, (__hip_bfloat16*)out,       \
                    hlast, B, Tseq, n_dir);                                    \
            }                                                                  \
        } else {                                                               \
            if (cfg.bt == 32) {                                                \
                auto k = gru_fwd_kernel<float, 32, HPV, false>;                \
                hipFuncSetAttribute((const void*)k,                            \
                    hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);     \
                k<<<grid, 256, lds, stream>>>((const float*)gi,                \
                    (const float*)w, bhh, (float*)out, hlast, B, Tseq, n_dir); \
            } else {                                                           \
                auto k = gru_fwd_kernel<float, 16, HPV, false>;                \
                hipFuncSetAttribute((const void*)k,                            \
                    hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);     \
                k<<<grid, 256, lds, stream>>>((const float*)gi,                \
                    (const float*)w, bhh, (float*)out, hlast, B, Tseq, n_dir); \
            }                                                                  \
        }                                                                      \
    } while (0)

    FMDA_DISPATCH_HP(Hp, FWD_CASE)
#undef FWD_CASE
    err = hipGetLastError();
    return err == hipSuccess ? 0 : (int)err;
}

extern "C" int fmda_gru_bwd_launch(int is_bf16, int Hp, const void* gi,
                                   const void* w, const float* bhh,
                                   const void* out, const void* dout,
                                   const float* dhT, void* dgi, void* dgh,
                                   float* dh0, float* dbhh, int B, int Tseq,
                                   int n_dir, hipStream_t stream) {
    if (!is_bf16 && Hp > 256) return -3;  // fp32 oracle unsupported at H=512
    const LaunchCfg cfg = bwd_cfg(is_bf16, Hp);
    const size_t lds = bwd_lds_bytes(is_bf16, Hp, cfg.bt, cfg.wlds);
    if (lds > 160 * 1024) return -2;
    const dim3 grid((B + cfg.bt - 1) / cfg.bt, n_dir);
    hipError_t err = hipSuccess;

#define BWD_KERNEL(TY, BTV, HPV, WL)                                           \
    do {                                                                       \
        auto k = gru_bwd_kernel<TY, BTV, HPV, WL>;                             \
        hipFuncSetAttribute((const void*)k,                                    \
            hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);             \
        k<<<grid, 256, lds, stream>>>((const TY*)gi, (const TY*)w, bhh,        \
            (const TY*)out, (const TY*)dout, dhT, (TY*)dgi, (TY*)dgh, dh0,     \
            dbhh, B, Tseq, n_dir);                                             \
    } while (0)

#define BWD_CASE(HPV)                                                          \
    do {                                                                       \
        if (is_bf16) {                                                         \
            if (cfg.bt == 32 && cfg.wlds)                                      \
                BWD_KERNEL(__hip_bfloat16, 32, HPV, true);                     \
            else if (cfg.bt == 16 && cfg.wlds)                                 \
                BWD_KERNEL(__hip_bfloat16, 16, HPV, true);                     \
            else if (cfg.bt == 32)                                             \
                BWD_KERNEL(__hip_bfloat16, 32, HPV, false);                    \
            else                                                               \
                BWD_KERNEL(__hip_bfloat16, 16, HPV, false);                    \
        } else {                                                               \
            if (cfg.bt == 32) BWD_KERNEL(float, 32, HPV, false);               \
            else BWD_KERNEL(float, 16, HPV, false);                            \
        }                                                                      \
    } while (0)

    FMDA_DISPATCH_HP(Hp, BWD_CASE)
#undef BWD_CASE
#undef BWD_KERNEL
    err = hipGetLastError();
    return err == hipSuccess ? 0 : (int)err;
}

extern "C" int fmda_mfma_selftest_launch(const void* A, const void* Bm,
                                         float* C, hipStream_t stream) {
    mfma_selftest_kernel<<<1, 64, 0, stream>>>((const __hip_bfloat16*)A,
                                               (const __hip_bfloat16*)Bm, C);
    return hipGetLastError() == hipSuccess ? 0 : 1;
}

}  // namespace fmda
