// handyrl_amd CDNA4 (gfx950) kernels + torch bindings.
//
// Native HIP, written for MI355X: wave64, one launch per logical op.
// Components:
//   * target_scan     — fused backward scans for TD(lambda) / UPGO / V-Trace
//                       (semantics of reference handyrl/losses.py:20-60, one
//                       kernel instead of ~3*T tiny launches per call)
//   * masked_sample   — batched legal-action softmax sampling for GPU actors
//                       (semantics of reference handyrl/generation.py:53-60)

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

#define DEV_INLINE __device__ __forceinline__

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

DEV_INLINE float bf2f(short s) {
    unsigned u = ((unsigned)(unsigned short)s) << 16;
    return __builtin_bit_cast(float, u);
}

DEV_INLINE short f2bf(float f) {   // round-to-nearest-even
    unsigned u = __builtin_bit_cast(unsigned, f);
    u += 0x7fffu + ((u >> 16) & 1u);
    return (short)(u >> 16);
}

enum ScanKind { KIND_TD = 0, KIND_UPGO = 1, KIND_VTRACE = 2 };

// One lane scans one (b, p) trajectory serially over T.  Work per call is
// tiny (B*P lanes x T steps); the win over eager is launch-count, not FLOPs.
__global__ void target_scan_kernel(
    const float* __restrict__ values,    // (B,T,P)
    const float* __restrict__ ret_boot,  // (B,P): bootstrap value (returns at T-1)
    const float* __restrict__ rewards,   // (B,T,P) or nullptr
    const float* __restrict__ lambda_,   // (B,T,P)
    const float* __restrict__ rhos,      // (B,T,P), VTRACE only
    const float* __restrict__ cs,        // (B,T,P), VTRACE only
    float* __restrict__ targets,
    float* __restrict__ adv,
    int n_bp, int T, int P, float gamma, int kind)
{
    const int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n_bp) return;
    const int b = idx / P, p = idx % P;
    const long base = (long)b * T * P + p;
    const long stride = P;

#define AT(a, t) a[base + (long)(t) * stride]

    const float ret_last = ret_boot[(long)b * P + p];

    if (kind == KIND_VTRACE) {
        // vs(t) = V(t) + sum-scan of deltas; advantage uses vs(t+1)
        float carry = 0.f;           // vs_minus_v at t+1
        float vs_next = ret_last;    // vs at t+1 (bootstrap = final return)
        for (int t = T - 1; t >= 0; --t) {
            const float v = AT(values, t);
            const float r = rewards ? AT(rewards, t) : 0.f;
            const float v_next = (t == T - 1) ? ret_last : AT(values, t + 1);
            const float delta = AT(rhos, t) * (r + gamma * v_next - v);
            const float vmv = (t == T - 1)
                ? delta
                : delta + gamma * AT(lambda_, t + 1) * AT(cs, t) * carry;
            carry = vmv;
            const float vs = vmv + v;
            AT(targets, t) = vs;
            AT(adv, t) = r + gamma * vs_next - v;
            vs_next = vs;
        }
    } else {
        float tv = ret_last;
        AT(targets, T - 1) = tv;
        AT(adv, T - 1) = tv - AT(values, T - 1);
        for (int t = T - 2; t >= 0; --t) {
            const float v1 = AT(values, t + 1);
            const float r = rewards ? AT(rewards, t) : 0.f;
            const float lam = AT(lambda_, t + 1);
            float mix = (1.f - lam) * v1 + lam * tv;
            if (kind == KIND_UPGO) mix = fmaxf(v1, mix);
            tv = r + gamma * mix;
            AT(targets, t) = tv;
            AT(adv, t) = tv - AT(values, t);
        }
    }
#undef AT
}

// One lane per row; A <= a few hundred, all L2-resident.  Three passes in
// registers: row max, exp-sum, then inverse-CDF selection at u * sum.
__global__ void masked_sample_kernel(
    const float* __restrict__ logits,   // (N, A)
    const float* __restrict__ mask,     // (N, A), additive: 0 legal / 1e32 illegal
    const float* __restrict__ uniform,  // (N,) in [0, 1)
    long* __restrict__ action,
    float* __restrict__ prob,
    int N, int A)
{
    const int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= N) return;
    const float* lg = logits + (long)i * A;
    const float* mk = mask + (long)i * A;

    float m = -INFINITY;
    for (int a = 0; a < A; ++a) m = fmaxf(m, lg[a] - mk[a]);
    float s = 0.f;
    for (int a = 0; a < A; ++a) s += __expf(lg[a] - mk[a] - m);

    const float u = uniform[i] * s;
    float acc = 0.f;
    int sel = -1;
    float pr = 0.f;
    for (int a = 0; a < A; ++a) {
        const float e = __expf(lg[a] - mk[a] - m);
        acc += e;
        if (sel < 0 && u < acc) { sel = a; pr = e / s; }
    }
    if (sel < 0) {   // numerical tail: last legal action
        for (int a = A - 1; a >= 0; --a) {
            if (mk[a] == 0.f) {
                sel = a;
                pr = __expf(lg[a] - mk[a] - m) / s;
                break;
            }
        }
        if (sel < 0) sel = 0;
    }
    action[i] = sel;
    prob[i] = pr;
}

// --- BatchNorm2d training fwd/bwd ---------------------------------------
// nn.BatchNorm2d semantics (biased var for normalization, unbiased for the
// running-var update).  Layout (N,C,HW): a "row" = one (n,c) slice of HW
// contiguous elements; rows are contiguous in memory (flat row id =
// n*C + c).  Stats kernels: grid C*S with one whole row per thread (lines
// stay L1-resident across the scalar sweep) + global fp32 atomics; apply
// kernels: one row per thread.  3 graph nodes per direction, replacing the
// ~22-kernel torch-primitive composition per layer whose strided reduces
// (40-85us each) dominated the captured train step.

template <typename T>
DEV_INLINE float load_as_f32(const T* p);
template <> DEV_INLINE float load_as_f32<short>(const short* p) { return bf2f(*p); }
template <> DEV_INLINE float load_as_f32<float>(const float* p) { return *p; }
template <typename T>
DEV_INLINE void store_f32(T* p, float v);
template <> DEV_INLINE void store_f32<short>(short* p, float v) { *p = f2bf(v); }
template <> DEV_INLINE void store_f32<float>(float* p, float v) { *p = v; }

template <typename T>
__global__ __launch_bounds__(256) void bn_stats_kernel(
    const T* __restrict__ x, float* __restrict__ accum,  // (2C): sum, sumsq
    int N, int C, int HW, int S)
{
    const int c = blockIdx.x % C;
    const int slice = blockIdx.x / C;
    const int rows_per = (N + S - 1) / S;
    const int n0 = slice * rows_per;
    const int n1 = min(N, n0 + rows_per);
    float acc = 0.f, acc2 = 0.f;
    for (int n = n0 + (int)threadIdx.x; n < n1; n += 256) {
        const T* row = x + ((long)n * C + c) * HW;
        for (int i = 0; i < HW; ++i) {
            const float v = load_as_f32(row + i);
            acc += v;
            acc2 += v * v;
        }
    }
    __shared__ float s1[256], s2[256];
    s1[threadIdx.x] = acc; s2[threadIdx.x] = acc2;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {
        if ((int)threadIdx.x < s) {
            s1[threadIdx.x] += s1[threadIdx.x + s];
            s2[threadIdx.x] += s2[threadIdx.x + s];
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        atomicAdd(&accum[c], s1[0]);
        atomicAdd(&accum[C + c], s2[0]);
    }
}

template <typename T>
__global__ __launch_bounds__(256) void bn_apply_kernel(
    const T* __restrict__ x, T* __restrict__ y,
    const float* __restrict__ accum,
    const float* __restrict__ weight, const float* __restrict__ bias,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    float* __restrict__ save_mean, float* __restrict__ save_rstd,
    int N, int C, int HW, float momentum, float eps)
{
    const long row = (long)blockIdx.x * 256 + threadIdx.x;   // over N*C
    const float M = (float)N * HW;
    if (row < C) {   // one thread per channel also publishes the stats
        const int c = (int)row;
        const float mean = accum[c] / M;
        const float var = fmaxf(accum[C + c] / M - mean * mean, 0.f);
        save_mean[c] = mean;
        save_rstd[c] = rsqrtf(var + eps);
        const float unbiased = var * (M / fmaxf(M - 1.f, 1.f));
        running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
        running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
    if (row >= (long)N * C) return;
    const int c = (int)(row % C);
    const float mean = accum[c] / M;
    const float var = fmaxf(accum[C + c] / M - mean * mean, 0.f);
    const float rstd = rsqrtf(var + eps);
    const float scale = weight[c] * rstd;
    const float shift = bias[c] - mean * scale;
    const T* xr = x + row * HW;
    T* yr = y + row * HW;
    for (int i = 0; i < HW; ++i)
        store_f32(yr + i, load_as_f32(xr + i) * scale + shift);
}

template <typename T>
__global__ __launch_bounds__(256) void bn_bwd_stats_kernel(
    const T* __restrict__ x, const T* __restrict__ dy,
    float* __restrict__ accum,   // (2C): sum dy, sum dy*xhat
    const float* __restrict__ save_mean, const float* __restrict__ save_rstd,
    int N, int C, int HW, int S)
{
    const int c = blockIdx.x % C;
    const int slice = blockIdx.x / C;
    const int rows_per = (N + S - 1) / S;
    const int n0 = slice * rows_per;
    const int n1 = min(N, n0 + rows_per);
    const float mean = save_mean[c], rstd = save_rstd[c];
    float acc = 0.f, acc2 = 0.f;
    for (int n = n0 + (int)threadIdx.x; n < n1; n += 256) {
        const long off = ((long)n * C + c) * HW;
        for (int i = 0; i < HW; ++i) {
            const float g = load_as_f32(dy + off + i);
            const float xh = (load_as_f32(x + off + i) - mean) * rstd;
            acc += g;
            acc2 += g * xh;
        }
    }
    __shared__ float s1[256], s2[256];
    s1[threadIdx.x] = acc; s2[threadIdx.x] = acc2;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {
        if ((int)threadIdx.x < s) {
            s1[threadIdx.x] += s1[threadIdx.x + s];
            s2[threadIdx.x] += s2[threadIdx.x + s];
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        atomicAdd(&accum[c], s1[0]);
        atomicAdd(&accum[C + c], s2[0]);
    }
}

template <typename T>
__global__ __launch_bounds__(256) void bn_bwd_apply_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, T* __restrict__ dx,
    const float* __restrict__ accum,
    const float* __restrict__ weight,
    const float* __restrict__ save_mean, const float* __restrict__ save_rstd,
    float* __restrict__ dweight, float* __restrict__ dbias,
    int N, int C, int HW)
{
    const long row = (long)blockIdx.x * 256 + threadIdx.x;
    const float M = (float)N * HW;
    if (row < C) {
        dbias[row] = accum[row];
        dweight[row] = accum[C + row];
    }
    if (row >= (long)N * C) return;
    const int c = (int)(row % C);
    const float mean = save_mean[c], rstd = save_rstd[c];
    const float k_dy = accum[c] / M;
    const float k_dyx = accum[C + c] / M;
    const float wr = weight[c] * rstd;
    const long off = row * HW;
    for (int i = 0; i < HW; ++i) {
        const float g = load_as_f32(dy + off + i);
        const float xh = (load_as_f32(x + off + i) - mean) * rstd;
        store_f32(dx + off + i, wr * (g - k_dy - xh * k_dyx));
    }
}

// --- weight packing for the MFMA torus conv ------------------------------
// Packs (co=32, ci<=32, 3, 3) fp32 conv weights into the B-fragment layout
// frag[tap][cotile][khi][lane_lo][e] (bf16), optionally scaled per out-
// channel (BN fold) and optionally in DGRAD form (flip taps, swap ci/co).
// One kernel instead of ~10 torch ops per layer per refresh.
__global__ void pack_torus_weights_kernel(
    const float* __restrict__ w,      // (32, ci_in, 3, 3)
    const float* __restrict__ scale,  // (32,) or nullptr
    short* __restrict__ frag,         // (9, 2, 4, 16, 8) bf16
    int ci_in, int dgrad)
{
    // flat output index over 9*2*4*16*8 = 9216
    const int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= 9 * 2 * 4 * 16 * 8) return;
    const int e = idx & 7;
    const int lo = (idx >> 3) & 15;
    const int khi = (idx >> 7) & 3;
    const int cotile = (idx >> 9) & 1;
    const int tap = idx >> 10;
    const int k = khi * 8 + e;            // ci index (0..31)
    const int n = cotile * 16 + lo;       // co index (0..31)
    float v = 0.f;
    if (!dgrad) {
        // frag[k=ci][n=co] at tap
        if (k < ci_in)
            v = w[((n * ci_in) + k) * 9 + tap];
        if (scale) v *= scale[n];
    } else {
        // dgrad conv: W'[tap][k=co][n=ci] = W[co=k][ci=n][8-tap]
        if (n < ci_in)
            v = w[((k * ci_in) + n) * 9 + (8 - tap)];
    }
    frag[idx] = f2bf(v);
}

// --- NHWC BatchNorm (training) with fused residual-add + ReLU ------------
// Activations (N, 77, 32) bf16: channels are innermost, so thread t
// (stride 256, 256 % 32 == 0) always sees channel c = t & 31 — private
// accumulators + perfectly coalesced loads.  apply: y = relu(bn(x) + res).
template <typename T>
__global__ __launch_bounds__(1024) void bn_nhwc_stats_kernel(
    const T* __restrict__ x, float* __restrict__ accum,   // (2C)
    long total, int C)
{
    const int c = threadIdx.x & 31;
    float acc = 0.f, acc2 = 0.f;
    for (long i = (long)blockIdx.x * 1024 + threadIdx.x; i < total;
         i += (long)gridDim.x * 1024) {
        const float v = load_as_f32(x + i);
        acc += v;
        acc2 += v * v;
    }
    __shared__ float s1[1024], s2[1024];
    s1[threadIdx.x] = acc; s2[threadIdx.x] = acc2;
    __syncthreads();
    // reduce the 32 threads per channel (stride 32 within the block)
    if (threadIdx.x < 32) {
        float t1 = 0.f, t2 = 0.f;
        for (int j = threadIdx.x; j < 1024; j += 32) { t1 += s1[j]; t2 += s2[j]; }
        atomicAdd(&accum[c], t1);
        atomicAdd(&accum[C + c], t2);
    }
}

template <typename T>
__global__ __launch_bounds__(256) void bn_nhwc_apply_kernel(
    const T* __restrict__ x, const T* __restrict__ res,   // res nullable
    T* __restrict__ y,
    const float* __restrict__ accum,
    const float* __restrict__ weight, const float* __restrict__ bias,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    float* __restrict__ save_mean, float* __restrict__ save_rstd,
    long total, int C, float momentum, float eps, int apply_relu)
{
    const long i0 = (long)blockIdx.x * 256 + threadIdx.x;
    const float M = (float)(total / C);
    if (i0 < C) {    // publish stats + running updates once
        const int c = (int)i0;
        const float mean = accum[c] / M;
        const float var = fmaxf(accum[C + c] / M - mean * mean, 0.f);
        save_mean[c] = mean;
        save_rstd[c] = rsqrtf(var + eps);
        const float unbiased = var * (M / fmaxf(M - 1.f, 1.f));
        running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
        running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
    const int c = threadIdx.x & 31;
    const float mean = accum[c] / M;
    const float var = fmaxf(accum[C + c] / M - mean * mean, 0.f);
    const float rstd = rsqrtf(var + eps);
    const float scale = weight[c] * rstd;
    const float shift = bias[c] - mean * scale;
    for (long i = i0; i < total; i += (long)gridDim.x * 256) {
        float v = load_as_f32(x + i) * scale + shift;
        if (res) v += load_as_f32(res + i);
        if (apply_relu) v = fmaxf(v, 0.f);
        store_f32(y + i, v);
    }
}

// backward of y = relu(bn(x) + res): given dy and y, the relu mask is
// y > 0; dz = dy * mask flows to BOTH the bn input grad and the residual
// grad (dres = dz).  stats: sum(dz), sum(dz * xhat).
template <typename T>
__global__ __launch_bounds__(1024) void bn_nhwc_bwd_stats_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, const T* __restrict__ y,
    float* __restrict__ accum,
    const float* __restrict__ save_mean, const float* __restrict__ save_rstd,
    long total, int C, int had_relu)
{
    const int c = threadIdx.x & 31;
    const float mean = save_mean[c], rstd = save_rstd[c];
    float acc = 0.f, acc2 = 0.f;
    for (long i = (long)blockIdx.x * 1024 + threadIdx.x; i < total;
         i += (long)gridDim.x * 1024) {
        float g = load_as_f32(dy + i);
        if (had_relu && load_as_f32(y + i) <= 0.f) g = 0.f;
        const float xh = (load_as_f32(x + i) - mean) * rstd;
        acc += g;
        acc2 += g * xh;
    }
    __shared__ float s1[1024], s2[1024];
    s1[threadIdx.x] = acc; s2[threadIdx.x] = acc2;
    __syncthreads();
    if (threadIdx.x < 32) {
        float t1 = 0.f, t2 = 0.f;
        for (int j = threadIdx.x; j < 1024; j += 32) { t1 += s1[j]; t2 += s2[j]; }
        atomicAdd(&accum[c], t1);
        atomicAdd(&accum[C + c], t2);
    }
}

template <typename T>
__global__ __launch_bounds__(256) void bn_nhwc_bwd_apply_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, const T* __restrict__ y,
    T* __restrict__ dx, T* __restrict__ dres,            // dres nullable
    const float* __restrict__ accum,
    const float* __restrict__ weight,
    const float* __restrict__ save_mean, const float* __restrict__ save_rstd,
    float* __restrict__ dweight, float* __restrict__ dbias,
    long total, int C, int had_relu)
{
    const long i0 = (long)blockIdx.x * 256 + threadIdx.x;
    if (i0 < C) {
        dbias[i0] = accum[i0];
        dweight[i0] = accum[C + i0];
    }
    const int c = threadIdx.x & 31;
    const float M = (float)(total / C);
    const float mean = save_mean[c], rstd = save_rstd[c];
    const float k_dy = accum[c] / M;
    const float k_dyx = accum[C + c] / M;
    const float wr = weight[c] * rstd;
    for (long i = i0; i < total; i += (long)gridDim.x * 256) {
        float g = load_as_f32(dy + i);
        if (had_relu && load_as_f32(y + i) <= 0.f) g = 0.f;
        if (dres) store_f32(dres + i, g);
        const float xh = (load_as_f32(x + i) - mean) * rstd;
        store_f32(dx + i, wr * (g - k_dy - xh * k_dyx));
    }
}

// --- torus conv weight gradient on MFMA ----------------------------------
// dW[tap][ci][co] = sum_pos x[pos][ci] * dy[nbr(pos, 8-tap)][co]
// (reindexed so the x^T A-operand is SHARED by all 9 taps).  One image
// (77 positions + zero padding to 96) is staged in LDS per iteration; each
// of the 4 waves owns one 16x16 (ci, co) quadrant for ALL 9 taps
// (9 f32x4 accumulators) and runs 3 k-steps x 9 taps of
// mfma_f32_16x16x32_bf16 per image.  Partials land in fp32 dW via one
// atomicAdd per element per block.  Replaces an 87 MB index_select gather
// + strided bmm per layer (3.6 ms + 1.7 ms per captured step).
__global__ __launch_bounds__(256) void torus_wgrad_kernel(
    const short* __restrict__ x,     // (N,77,32) bf16
    const short* __restrict__ dy,    // (N,77,32) bf16
    const int* __restrict__ nbr,     // (77,9)
    float* __restrict__ dW,          // (9,32,32) fp32, pre-zeroed
    int N)
{
    __shared__ short x_lds[96 * 32];
    __shared__ short dy_lds[96 * 32];
    __shared__ int nbr_inv[96 * 9];    // nbr_inv[pos][tap] = nbr[pos][8-tap]

    const int tid = threadIdx.x;
    const int wid = tid >> 6, lane = tid & 63;
    const int khi = lane >> 4, lo = lane & 15;
    const int citile = wid >> 1, cotile = wid & 1;

    // inverse-neighbor table: positions >= 77 point at the zero row 95
    for (int i = tid; i < 96 * 9; i += 256) {
        const int pos = i / 9, tap = i % 9;
        nbr_inv[i] = (pos < 77) ? nbr[pos * 9 + (8 - tap)] : 95;
    }
    // zero the padding rows once (rows 77..95 never rewritten)
    for (int i = tid; i < (96 - 77) * 32; i += 256) {
        x_lds[77 * 32 + i] = 0;
        dy_lds[77 * 32 + i] = 0;
    }
    __syncthreads();

    f32x4 acc[9];
#pragma unroll
    for (int t = 0; t < 9; ++t) acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int n = blockIdx.x; n < N; n += gridDim.x) {
        // stage one image of x and dy (2464 bf16 each, contiguous)
        const short* xs = x + (long)n * 77 * 32;
        const short* ds = dy + (long)n * 77 * 32;
        for (int i = tid; i < 77 * 32 / 4; i += 256) {
            ((short4*)x_lds)[i] = ((const short4*)xs)[i];
            ((short4*)dy_lds)[i] = ((const short4*)ds)[i];
        }
        __syncthreads();

#pragma unroll
        for (int ks = 0; ks < 3; ++ks) {
            // A fragment: x^T, m=ci, k=pos (shared by all taps)
            bf16x8 a;
            const int ci = citile * 16 + lo;
#pragma unroll
            for (int i = 0; i < 8; ++i) {
                const int pos = ks * 32 + khi * 8 + i;
                a[i] = __builtin_bit_cast(__bf16, x_lds[pos * 32 + ci]);
            }
            const int co = cotile * 16 + lo;
#pragma unroll
            for (int t = 0; t < 9; ++t) {
                bf16x8 b;
#pragma unroll
                for (int i = 0; i < 8; ++i) {
                    const int pos = ks * 32 + khi * 8 + i;
                    b[i] = __builtin_bit_cast(
                        __bf16, dy_lds[nbr_inv[pos * 9 + t] * 32 + co]);
                }
                acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t],
                                                                 0, 0, 0);
            }
        }
        __syncthreads();
    }

    // one atomic partial-add per output element per block
#pragma unroll
    for (int t = 0; t < 9; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int ci = citile * 16 + khi * 4 + r;
            const int co = cotile * 16 + lo;
            atomicAdd(&dW[(t * 32 + ci) * 32 + co], acc[t][r]);
        }
    }
}

// --- MFMA fragment-layout probe (test harness for the conv kernel) ------
// Computes D(16x16) = A(16x32) @ B(32x16) with one v_mfma_f32_16x16x32_bf16
// using the assumed lane->fragment mapping:
//   A: lane l holds A[m = l&15][k = (l>>4)*8 + i], i in [0,8)
//   B: lane l holds B[k = (l>>4)*8 + i][n = l&15]
//   D: lane l, reg r -> D[row = (l>>4)*4 + r][col = l&15]
// Verified against torch.matmul on gfx950 by tests/test_gpu.py.
__global__ void mfma_probe_kernel(const short* __restrict__ A,
                                  const short* __restrict__ B,
                                  float* __restrict__ D) {
    const int l = threadIdx.x;
    const int khi = l >> 4, lo = l & 15;
    bf16x8 a, b;
    for (int i = 0; i < 8; ++i) {
        a[i] = __builtin_bit_cast(__bf16, A[lo * 32 + khi * 8 + i]);
        b[i] = __builtin_bit_cast(__bf16, B[(khi * 8 + i) * 16 + lo]);
    }
    f32x4 c = {0.f, 0.f, 0.f, 0.f};
    c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    for (int r = 0; r < 4; ++r)
        D[(khi * 4 + r) * 16 + lo] = c[r];
}

// --- obs (N,17,7,11) uint8 NCHW -> (N,77,32) bf16 NHWC, channels padded --
__global__ void obs_to_nhwc_kernel(const unsigned char* __restrict__ obs,
                                   short* __restrict__ out, long total) {
    const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;  // N*77*32
    if (idx >= total) return;
    const int c = idx & 31;
    const long pc = idx >> 5;
    const long n = pc / 77;
    const int cell = pc % 77;
    float v = (c < 17) ? (float)obs[(n * 17 + c) * 77 + cell] : 0.f;
    out[idx] = f2bf(v);
}

// --- canonical obs (N,17,77) u8 -> seat-expanded (N*4,77,32) bf16 NHWC --
// Output row n = g*4 + k is seat k's view of game g: channel c < 16 reads
// canonical channel (c & 12) | ((c + k) & 3) (the four goose planes of each
// group rotated so the seat's own goose comes first), channel 16 = food.
// Host code ships ONE board per game (4x less bytes end-to-end); the seat
// rotation is free address math here.
__global__ void obs_to_nhwc_rot_kernel(const unsigned char* __restrict__ obs,
                                       short* __restrict__ out, long total) {
    const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;  // N*4*77*32
    if (idx >= total) return;
    const int c = idx & 31;
    const long pc = idx >> 5;
    const long row = pc / 77;              // g*4 + k
    const int cell = pc % 77;
    const long g = row >> 2;
    const int k = row & 3;
    float v = 0.f;
    if (c < 17) {
        const int src = (c < 16) ? ((c & 12) | ((c + k) & 3)) : 16;
        v = (float)obs[(g * 17 + src) * 77 + cell];
    }
    out[idx] = f2bf(v);
}

// --- fused torus conv block: y = act(conv3x3_wrap(x)*scale + shift [+ x]) -
// Implicit GEMM on MFMA, NHWC bf16 activations, BN folded into the packed
// weights (wfrag) and the per-channel shift.  The 3x3 wrap-around gather is
// the A-operand load itself: k is tap-major (k = tap*32 + ci), so one
// A-fragment = 16 contiguous bytes of the neighbor cell's channel vector.
// Replaces the reference eval chain circular-pad + conv + BN + (+x) + relu
// (reference envs/kaggle/hungry_geese.py:30-35, 48-51) with ONE kernel.
//
// Geometry: 256 threads = 4 waves; each wave computes a 16-position x 32-
// channel output tile (2 MFMA accumulators), block covers 64 positions;
// 9 taps -> 18 MFMA per wave.  W (18 KB) streams from L2 (shared by all
// blocks); per-lane fragment rows are packed host-side so every read is a
// contiguous 16-byte load.
__global__ __launch_bounds__(256) void torus_conv_fused_kernel(
    const short* __restrict__ x,        // (N,77,32) bf16
    const short* __restrict__ wfrag,    // (9,2,4,16,8) bf16, BN-folded
    const float* __restrict__ shift,    // (32,)
    const int* __restrict__ nbr,        // (77,9) wrap-around neighbor cells
    const short* __restrict__ res,      // nullable residual, (N,77,32) bf16
    short* __restrict__ y,              // (N,77,32) bf16
    long total_pos,                     // N*77
    int apply_relu) {
    const int wid = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int khi = lane >> 4, lo = lane & 15;
    const long pos_base = (long)blockIdx.x * 64 + wid * 16;

    const long p = pos_base + lo;                  // this lane's A row
    const long pp = (p < total_pos) ? p : 0;
    const long n = pp / 77;
    const int cell = (int)(pp % 77);

    const bf16x8* wf = (const bf16x8*)wfrag;
    f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
    f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
        const int nb = nbr[cell * 9 + tap];
        const bf16x8 a = *(const bf16x8*)(x + (n * 77 + nb) * 32 + khi * 8);
        const bf16x8 b0 = wf[((tap * 2 + 0) * 4 + khi) * 16 + lo];
        const bf16x8 b1 = wf[((tap * 2 + 1) * 4 + khi) * 16 + lo];
        acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b0, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b1, acc1, 0, 0, 0);
    }

    const float sh0 = shift[lo], sh1 = shift[16 + lo];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const long prow = pos_base + khi * 4 + r;
        if (prow >= total_pos) continue;
        float v0 = acc0[r] + sh0;
        float v1 = acc1[r] + sh1;
        const long off = prow * 32;
        if (res) {
            v0 += bf2f(res[off + lo]);
            v1 += bf2f(res[off + 16 + lo]);
        }
        if (apply_relu) {
            v0 = fmaxf(v0, 0.f);
            v1 = fmaxf(v1, 0.f);
        }
        y[off + lo] = f2bf(v0);
        y[off + 16 + lo] = f2bf(v1);
    }
}

// --- fused loss head (flagship FF solo family) ----------------------------
// Replaces the ~40 eager elementwise/reduce ops of compute_loss around the
// target scans (reference train.py:229-267, 189-215) with three kernels:
//   pre : log-softmax gather, importance ratios, value/outcome splice
//   fwd : reduced loss sums (p, v, H, H*decay) + data count
//   bwd : d(policy logits), d(value) from the four loss-component grads
// One thread per (b, t) row; the action dimension is a register loop.

__global__ void loss_head_pre_kernel(
    const float* __restrict__ policy,   // (n, A) masked logits
    const long* __restrict__ action,    // (n,)
    const float* __restrict__ mu,       // (n,) behavior prob
    const float* __restrict__ emask,    // (n,)
    const float* __restrict__ value,    // (n,)
    const float* __restrict__ outcome,  // (n,)
    float* __restrict__ log_sel,        // (n,) out
    float* __restrict__ rho_clip,       // (n,) out (clip 1.0; == c)
    float* __restrict__ v_spliced,      // (n,) out
    long n, int A)
{
    const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const float* row = policy + i * A;
    float mx = row[0];
    for (int j = 1; j < A; ++j) mx = fmaxf(mx, row[j]);
    float se = 0.f;
    for (int j = 0; j < A; ++j) se += __expf(row[j] - mx);
    const float lse = mx + __logf(se);
    const float em = emask[i];
    const float ls = (row[action[i]] - lse) * em;
    log_sel[i] = ls;
    const float lb = __logf(fminf(fmaxf(mu[i], 1e-16f), 1.f)) * em;
    rho_clip[i] = fminf(__expf(ls - lb), 1.f);
    v_spliced[i] = value[i] * em + outcome[i] * (1.f - em);
}

__global__ void loss_head_fwd_kernel(
    const float* __restrict__ policy,   // (n, A)
    const float* __restrict__ log_sel,  // (n,)
    const float* __restrict__ ta,       // (n,) total advantage (rho*adv)
    const float* __restrict__ tmask,    // (n,)
    const float* __restrict__ omask,    // (n,)
    const float* __restrict__ value,    // (n,)
    const float* __restrict__ vtarget,  // (n,)
    const float* __restrict__ progress, // (n,)
    float* __restrict__ accum,          // (5,): p, v, ent, ent_decay, dcnt
    float ent_decay, long n, int A)
{
    const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    float p_l = 0.f, v_l = 0.f, h_l = 0.f, hd_l = 0.f, dc = 0.f;
    if (i < n) {
        const float tm = tmask[i];
        p_l = -log_sel[i] * ta[i] * tm;
        const float dv = value[i] - vtarget[i];
        v_l = 0.5f * dv * dv * omask[i];
        // entropy of the masked logits
        const float* row = policy + i * A;
        float mx = row[0];
        for (int j = 1; j < A; ++j) mx = fmaxf(mx, row[j]);
        float se = 0.f;
        for (int j = 0; j < A; ++j) se += __expf(row[j] - mx);
        const float lse = mx + __logf(se);
        float H = 0.f;
        for (int j = 0; j < A; ++j) {
            const float lp = row[j] - lse;
            H -= __expf(lp) * lp;
        }
        h_l = H * tm;
        hd_l = h_l * (1.f - progress[i] * (1.f - ent_decay));
        dc = tm;
    }
    // block reduce + atomics
    __shared__ float s[256][5];
    s[threadIdx.x][0] = p_l; s[threadIdx.x][1] = v_l; s[threadIdx.x][2] = h_l;
    s[threadIdx.x][3] = hd_l; s[threadIdx.x][4] = dc;
    __syncthreads();
    for (int step = 128; step > 0; step >>= 1) {
        if ((int)threadIdx.x < step)
            for (int k = 0; k < 5; ++k)
                s[threadIdx.x][k] += s[threadIdx.x + step][k];
        __syncthreads();
    }
    if (threadIdx.x == 0)
        for (int k = 0; k < 5; ++k) atomicAdd(&accum[k], s[0][k]);
}

__global__ void loss_head_bwd_kernel(
    const float* __restrict__ policy,   // (n, A)
    const long* __restrict__ action,    // (n,)
    const float* __restrict__ ta,       // (n,)
    const float* __restrict__ tmask,    // (n,)
    const float* __restrict__ omask,    // (n,)
    const float* __restrict__ emask,    // (n,)
    const float* __restrict__ value,    // (n,)
    const float* __restrict__ vtarget,  // (n,)
    const float* __restrict__ progress, // (n,)
    const float* __restrict__ g,        // (5,): d/d accum[k] (see fwd)
    float* __restrict__ dpolicy,        // (n, A) out
    float* __restrict__ dvalue,         // (n,) out
    float ent_decay, long n, int A)
{
    const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    const float gp = g[0];
    const float gv = g[1];
    const float tm = tmask[i];
    dvalue[i] = gv * (value[i] - vtarget[i]) * omask[i];

    const float* row = policy + i * A;
    float mx = row[0];
    for (int j = 1; j < A; ++j) mx = fmaxf(mx, row[j]);
    float se = 0.f;
    for (int j = 0; j < A; ++j) se += __expf(row[j] - mx);
    const float lse = mx + __logf(se);
    float H = 0.f;
    for (int j = 0; j < A; ++j) {
        const float lp = row[j] - lse;
        H -= __expf(lp) * lp;
    }
    const float w_pg = gp * ta[i] * tm * emask[i];
    const float decay = 1.f - progress[i] * (1.f - ent_decay);
    const float e_w = (g[2] + g[3] * decay) * tm;
    const int a = (int)action[i];
    float* drow = dpolicy + i * A;
    for (int j = 0; j < A; ++j) {
        const float lp = row[j] - lse;
        const float pj = __expf(lp);
        float d = w_pg * (pj - (j == a ? 1.f : 0.f));
        d += e_w * (-pj * (lp + H));
        drow[j] = d;
    }
}

// --- fused ConvLSTM cell (Geister DRC core) -------------------------------
// One kernel per cell evaluation: implicit-GEMM 3x3 zero-pad conv over the
// K-ordered (x | h) input halves (concat never materialized) with the
// 4-gate sigmoid/tanh state update fused into the epilogue.  Replaces the
// reference cell's concat + conv + split + 8 pointwise ops
// (reference envs/geister.py:18-58) with ONE launch.
//
// Shapes: x, h (B, 36, 32) NHWC bf16; c (B, 36, 32) fp32; gates N = 128
// (i | f | o | g quarters, reference split order geister.py:46).
// Geometry: 256 threads = 4 waves; each wave computes a 16-position x
// 128-channel tile (8 MFMA accumulators); K = 2 halves x 9 taps x 32
// channels = 18 MFMA K-steps, 144 MFMA per wave.  The gate quartets land
// in the SAME lane (channel hc sits in acc tiles {hc/16, +2, +4, +6} at
// lane lo = hc % 16), so the state update needs no cross-lane traffic.
__global__ __launch_bounds__(256) void convlstm_cell_kernel(
    const short* __restrict__ x,        // (B*36, 32) bf16
    const short* __restrict__ h,        // (B*36, 32) bf16
    const float* __restrict__ c,        // (B*36, 32) fp32
    const short* __restrict__ wfrag,    // (2, 9, 8, 4, 16, 8) bf16
    const float* __restrict__ bias,     // (128,)
    const int* __restrict__ nbr,        // (36, 9) zero-pad: -1 = outside
    short* __restrict__ h_out,          // (B*36, 32) bf16
    float* __restrict__ c_out,          // (B*36, 32) fp32
    long total_pos)                     // B*36
{
    const int wid = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int khi = lane >> 4, lo = lane & 15;
    const long pos_base = (long)blockIdx.x * 64 + wid * 16;

    const long p = pos_base + lo;                  // this lane's A row
    const long pp = (p < total_pos) ? p : 0;
    const long n = pp / 36;
    const int cell = (int)(pp % 36);

    const bf16x8* wf = (const bf16x8*)wfrag;
    const bf16x8 azero = {};
    f32x4 acc[8];
#pragma unroll
    for (int t = 0; t < 8; ++t) acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

#pragma unroll
    for (int src = 0; src < 2; ++src) {
        const short* in = src == 0 ? x : h;
#pragma unroll
        for (int tap = 0; tap < 9; ++tap) {
            const int nb = nbr[cell * 9 + tap];
            const bf16x8 a = (nb >= 0)
                ? *(const bf16x8*)(in + (n * 36 + nb) * 32 + khi * 8)
                : azero;
            const bf16x8* wrow = wf + ((src * 9 + tap) * 8) * 64 + khi * 16 + lo;
#pragma unroll
            for (int t = 0; t < 8; ++t)
                acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a, wrow[t * 64], acc[t], 0, 0, 0);
        }
    }

    float b_i0 = bias[lo],       b_i1 = bias[16 + lo];
    float b_f0 = bias[32 + lo],  b_f1 = bias[48 + lo];
    float b_o0 = bias[64 + lo],  b_o1 = bias[80 + lo];
    float b_g0 = bias[96 + lo],  b_g1 = bias[112 + lo];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const long prow = pos_base + khi * 4 + r;
        if (prow >= total_pos) continue;
        const long off = prow * 32;
        // hidden channel lo
        {
            const float gi = 1.f / (1.f + __expf(-(acc[0][r] + b_i0)));
            const float gf = 1.f / (1.f + __expf(-(acc[2][r] + b_f0)));
            const float go = 1.f / (1.f + __expf(-(acc[4][r] + b_o0)));
            const float gg = tanhf(acc[6][r] + b_g0);
            const float cn = gf * c[off + lo] + gi * gg;
            c_out[off + lo] = cn;
            h_out[off + lo] = f2bf(go * tanhf(cn));
        }
        // hidden channel 16 + lo
        {
            const float gi = 1.f / (1.f + __expf(-(acc[1][r] + b_i1)));
            const float gf = 1.f / (1.f + __expf(-(acc[3][r] + b_f1)));
            const float go = 1.f / (1.f + __expf(-(acc[5][r] + b_o1)));
            const float gg = tanhf(acc[7][r] + b_g1);
            const float cn = gf * c[off + 16 + lo] + gi * gg;
            c_out[off + 16 + lo] = cn;
            h_out[off + 16 + lo] = f2bf(go * tanhf(cn));
        }
    }
}

}  // namespace

template <typename T>
static void bn_fwd_launch(torch::Tensor& x, torch::Tensor& y, torch::Tensor& accum,
                          torch::Tensor& weight, torch::Tensor& bias,
                          torch::Tensor& rmean, torch::Tensor& rvar,
                          torch::Tensor& smean, torch::Tensor& srstd,
                          int N, int C, int HW, int S,
                          double momentum, double eps, hipStream_t stream) {
    hipLaunchKernelGGL(bn_stats_kernel<T>, dim3(C * S), dim3(256), 0, stream,
        (const T*)x.data_ptr(), accum.data_ptr<float>(), N, C, HW, S);
    const long rows = (long)N * C;
    hipLaunchKernelGGL(bn_apply_kernel<T>, dim3((rows + 255) / 256), dim3(256), 0, stream,
        (const T*)x.data_ptr(), (T*)y.data_ptr(), accum.data_ptr<float>(),
        weight.data_ptr<float>(), bias.data_ptr<float>(),
        rmean.data_ptr<float>(), rvar.data_ptr<float>(),
        smean.data_ptr<float>(), srstd.data_ptr<float>(),
        N, C, HW, (float)momentum, (float)eps);
}

static std::vector<torch::Tensor> bn_train_fwd(
    torch::Tensor x, torch::Tensor weight, torch::Tensor bias,
    torch::Tensor running_mean, torch::Tensor running_var,
    double momentum, double eps) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 4, "bn_train_fwd: (N,C,H,W) expected");
    const int N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
    const int S = std::min(32, std::max(1, N / 256));
    auto y = torch::empty_like(x);
    auto accum = torch::empty({2 * C}, weight.options());
    auto save_mean = torch::empty({C}, weight.options());
    auto save_rstd = torch::empty({C}, weight.options());
    auto stream = at::cuda::getCurrentCUDAStream();
    hipMemsetAsync(accum.data_ptr(), 0, 2 * C * sizeof(float), stream);
    if (x.scalar_type() == torch::kBFloat16) {
        bn_fwd_launch<short>(x, y, accum, weight, bias, running_mean, running_var,
                             save_mean, save_rstd, N, C, HW, S, momentum, eps, stream);
    } else {
        TORCH_CHECK(x.scalar_type() == torch::kFloat32, "bn: bf16/f32 only");
        bn_fwd_launch<float>(x, y, accum, weight, bias, running_mean, running_var,
                             save_mean, save_rstd, N, C, HW, S, momentum, eps, stream);
    }
    return {y, save_mean, save_rstd};
}

template <typename T>
static void bn_bwd_launch(torch::Tensor& x, torch::Tensor& dy, torch::Tensor& dx,
                          torch::Tensor& accum, torch::Tensor& weight,
                          torch::Tensor& smean, torch::Tensor& srstd,
                          torch::Tensor& dweight, torch::Tensor& dbias,
                          int N, int C, int HW, int S, hipStream_t stream) {
    hipLaunchKernelGGL(bn_bwd_stats_kernel<T>, dim3(C * S), dim3(256), 0, stream,
        (const T*)x.data_ptr(), (const T*)dy.data_ptr(), accum.data_ptr<float>(),
        smean.data_ptr<float>(), srstd.data_ptr<float>(), N, C, HW, S);
    const long rows = (long)N * C;
    hipLaunchKernelGGL(bn_bwd_apply_kernel<T>, dim3((rows + 255) / 256), dim3(256), 0, stream,
        (const T*)x.data_ptr(), (const T*)dy.data_ptr(), (T*)dx.data_ptr(),
        accum.data_ptr<float>(), weight.data_ptr<float>(),
        smean.data_ptr<float>(), srstd.data_ptr<float>(),
        dweight.data_ptr<float>(), dbias.data_ptr<float>(), N, C, HW);
}

static std::vector<torch::Tensor> bn_train_bwd(
    torch::Tensor x, torch::Tensor dy, torch::Tensor weight,
    torch::Tensor save_mean, torch::Tensor save_rstd) {
    const int N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
    const int S = std::min(32, std::max(1, N / 256));
    auto dx = torch::empty_like(x);
    auto accum = torch::empty({2 * C}, weight.options());
    auto dweight = torch::empty({C}, weight.options());
    auto dbias = torch::empty({C}, weight.options());
    auto stream = at::cuda::getCurrentCUDAStream();
    hipMemsetAsync(accum.data_ptr(), 0, 2 * C * sizeof(float), stream);
    if (x.scalar_type() == torch::kBFloat16) {
        bn_bwd_launch<short>(x, dy, dx, accum, weight, save_mean, save_rstd,
                             dweight, dbias, N, C, HW, S, stream);
    } else {
        bn_bwd_launch<float>(x, dy, dx, accum, weight, save_mean, save_rstd,
                             dweight, dbias, N, C, HW, S, stream);
    }
    return {dx, dweight, dbias};
}

static torch::Tensor torus_wgrad(torch::Tensor x, torch::Tensor dy,
                                 torch::Tensor nbr) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(x.sizes() == dy.sizes() && x.size(1) == 77 && x.size(2) == 32);
    const int N = x.size(0);
    auto dW = torch::empty({9, 32, 32}, x.options().dtype(torch::kFloat32));
    auto stream = at::cuda::getCurrentCUDAStream();
    hipMemsetAsync(dW.data_ptr(), 0, 9 * 32 * 32 * sizeof(float), stream);
    const int blocks = std::max(1, std::min(N, 256));
    hipLaunchKernelGGL(torus_wgrad_kernel, dim3(blocks), dim3(256), 0, stream,
        (const short*)x.data_ptr(), (const short*)dy.data_ptr(),
        nbr.data_ptr<int>(), dW.data_ptr<float>(), N);
    return dW;
}

static torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
    TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16);
    auto D = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
        (const short*)A.data_ptr(), (const short*)B.data_ptr(),
        D.data_ptr<float>());
    return D;
}

static torch::Tensor obs_to_nhwc(torch::Tensor obs) {
    TORCH_CHECK(obs.is_cuda() && obs.scalar_type() == torch::kUInt8);
    const long N = obs.size(0);
    auto out = torch::empty({N, 77, 32}, obs.options().dtype(torch::kBFloat16));
    const long total = N * 77 * 32;
    const int block = 256;
    const long grid = (total + block - 1) / block;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(obs_to_nhwc_kernel, dim3(grid), dim3(block), 0, stream,
        obs.data_ptr<unsigned char>(), (short*)out.data_ptr(), total);
    return out;
}

static torch::Tensor obs_to_nhwc_rot(torch::Tensor obs) {
    TORCH_CHECK(obs.is_cuda() && obs.scalar_type() == torch::kUInt8);
    const long N = obs.size(0);
    auto out = torch::empty({N * 4, 77, 32},
                            obs.options().dtype(torch::kBFloat16));
    const long total = N * 4 * 77 * 32;
    const int block = 256;
    const long grid = (total + block - 1) / block;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(obs_to_nhwc_rot_kernel, dim3(grid), dim3(block), 0,
        stream, obs.data_ptr<unsigned char>(), (short*)out.data_ptr(), total);
    return out;
}

static torch::Tensor torus_conv_fused(
    torch::Tensor x, torch::Tensor wfrag, torch::Tensor shift,
    torch::Tensor nbr, c10::optional<torch::Tensor> res, bool apply_relu) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(x.dim() == 3 && x.size(1) == 77 && x.size(2) == 32,
                "torus_conv_fused: x must be (N,77,32) bf16");
    const long total_pos = x.size(0) * 77;
    auto y = torch::empty_like(x);
    const long grid = (total_pos + 63) / 64;
    auto stream = at::cuda::getCurrentCUDAStream();
    const short* res_ptr = res.has_value() ? (const short*)res->data_ptr() : nullptr;
    hipLaunchKernelGGL(torus_conv_fused_kernel, dim3(grid), dim3(256), 0, stream,
        (const short*)x.data_ptr(), (const short*)wfrag.data_ptr(),
        shift.data_ptr<float>(), nbr.data_ptr<int>(), res_ptr, (short*)y.data_ptr(),
        total_pos, (int)apply_relu);
    return y;
}

static std::vector<torch::Tensor> loss_head_pre(
    torch::Tensor policy, torch::Tensor action, torch::Tensor mu,
    torch::Tensor emask, torch::Tensor value, torch::Tensor outcome) {
    const long n = policy.size(0);
    const int A = policy.size(1);
    auto opts = policy.options();
    auto log_sel = torch::empty({n}, opts);
    auto rho = torch::empty({n}, opts);
    auto v_spl = torch::empty({n}, opts);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(loss_head_pre_kernel, dim3((n + 255) / 256), dim3(256),
        0, stream, policy.data_ptr<float>(), action.data_ptr<long>(),
        mu.data_ptr<float>(), emask.data_ptr<float>(),
        value.data_ptr<float>(), outcome.data_ptr<float>(),
        log_sel.data_ptr<float>(), rho.data_ptr<float>(),
        v_spl.data_ptr<float>(), n, A);
    return {log_sel, rho, v_spl};
}

static torch::Tensor loss_head_fwd(
    torch::Tensor policy, torch::Tensor log_sel, torch::Tensor ta,
    torch::Tensor tmask, torch::Tensor omask, torch::Tensor value,
    torch::Tensor vtarget, torch::Tensor progress, double ent_decay) {
    const long n = policy.size(0);
    const int A = policy.size(1);
    auto accum = torch::zeros({5}, policy.options());
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(loss_head_fwd_kernel, dim3((n + 255) / 256), dim3(256),
        0, stream, policy.data_ptr<float>(), log_sel.data_ptr<float>(),
        ta.data_ptr<float>(), tmask.data_ptr<float>(),
        omask.data_ptr<float>(), value.data_ptr<float>(),
        vtarget.data_ptr<float>(), progress.data_ptr<float>(),
        accum.data_ptr<float>(), (float)ent_decay, n, A);
    return accum;
}

static std::vector<torch::Tensor> loss_head_bwd(
    torch::Tensor policy, torch::Tensor action, torch::Tensor ta,
    torch::Tensor tmask, torch::Tensor omask, torch::Tensor emask,
    torch::Tensor value, torch::Tensor vtarget, torch::Tensor progress,
    torch::Tensor g, double ent_decay) {
    const long n = policy.size(0);
    const int A = policy.size(1);
    auto dpolicy = torch::empty_like(policy);
    auto dvalue = torch::empty({n}, policy.options());
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(loss_head_bwd_kernel, dim3((n + 255) / 256), dim3(256),
        0, stream, policy.data_ptr<float>(), action.data_ptr<long>(),
        ta.data_ptr<float>(), tmask.data_ptr<float>(),
        omask.data_ptr<float>(), emask.data_ptr<float>(),
        value.data_ptr<float>(), vtarget.data_ptr<float>(),
        progress.data_ptr<float>(), g.data_ptr<float>(),
        dpolicy.data_ptr<float>(), dvalue.data_ptr<float>(),
        (float)ent_decay, n, A);
    return {dpolicy, dvalue};
}

static std::vector<torch::Tensor> convlstm_cell(
    torch::Tensor x, torch::Tensor h, torch::Tensor c,
    torch::Tensor wfrag, torch::Tensor bias, torch::Tensor nbr,
    torch::Tensor h_out, torch::Tensor c_out) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
                x.dim() == 3 && x.size(1) == 36 && x.size(2) == 32,
                "convlstm_cell: x must be (B,36,32) bf16");
    TORCH_CHECK(h.sizes() == x.sizes() &&
                h.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(c.sizes() == x.sizes() &&
                c.scalar_type() == torch::kFloat32);
    TORCH_CHECK(wfrag.scalar_type() == torch::kBFloat16 &&
                wfrag.numel() == 2 * 9 * 8 * 4 * 16 * 8);
    const long total_pos = x.size(0) * 36;
    const long grid = (total_pos + 63) / 64;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(convlstm_cell_kernel, dim3(grid), dim3(256), 0, stream,
        (const short*)x.data_ptr(), (const short*)h.data_ptr(),
        c.data_ptr<float>(), (const short*)wfrag.data_ptr(),
        bias.data_ptr<float>(), nbr.data_ptr<int>(),
        (short*)h_out.data_ptr(), c_out.data_ptr<float>(), total_pos);
    return {h_out, c_out};
}

static torch::Tensor pack_torus_weights_hip(
    torch::Tensor w, c10::optional<torch::Tensor> scale, bool dgrad) {
    TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kFloat32 && w.dim() == 4);
    const int ci_in = w.size(1);
    TORCH_CHECK(!dgrad || ci_in == 32, "dgrad pack needs ci==32");
    auto frag = torch::empty({9, 2, 4, 16, 8}, w.options().dtype(torch::kBFloat16));
    auto stream = at::cuda::getCurrentCUDAStream();
    const float* sc = scale.has_value() ? scale->data_ptr<float>() : nullptr;
    hipLaunchKernelGGL(pack_torus_weights_kernel, dim3(36), dim3(256), 0, stream,
        w.data_ptr<float>(), sc, (short*)frag.data_ptr(), ci_in, (int)dgrad);
    return frag;
}

static int bn_grid(long total) {
    const long blocks = (total + 255) / 256;
    return (int)std::min<long>(blocks, 2048);
}

// stats kernels end with 64 atomics per block on 64 addresses: keep the
// block count low enough that contention stays off the critical path
static int bn_stats_grid(long total) {
    const long blocks = (total + 1023) / 1024;
    return (int)std::min<long>(blocks, 240);
}

static std::vector<torch::Tensor> bn_nhwc_fwd(
    torch::Tensor x, c10::optional<torch::Tensor> res,
    torch::Tensor weight, torch::Tensor bias,
    torch::Tensor running_mean, torch::Tensor running_var,
    double momentum, double eps, bool apply_relu) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.size(2) == 32,
                "bn_nhwc_fwd: (N,77,32) expected");
    const int C = x.size(2);
    const long total = x.numel();
    auto y = torch::empty_like(x);
    auto accum = torch::empty({2 * C}, weight.options());
    auto save_mean = torch::empty({C}, weight.options());
    auto save_rstd = torch::empty({C}, weight.options());
    auto stream = at::cuda::getCurrentCUDAStream();
    hipMemsetAsync(accum.data_ptr(), 0, 2 * C * sizeof(float), stream);
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "bn_nhwc: bf16 only");
    const short* res_ptr = res.has_value() ? (const short*)res->data_ptr() : nullptr;
    hipLaunchKernelGGL(bn_nhwc_stats_kernel<short>, dim3(bn_stats_grid(total)), dim3(1024), 0,
        stream, (const short*)x.data_ptr(), accum.data_ptr<float>(), total, C);
    hipLaunchKernelGGL(bn_nhwc_apply_kernel<short>, dim3(bn_grid(total)), dim3(256), 0,
        stream, (const short*)x.data_ptr(), res_ptr, (short*)y.data_ptr(),
        accum.data_ptr<float>(), weight.data_ptr<float>(), bias.data_ptr<float>(),
        running_mean.data_ptr<float>(), running_var.data_ptr<float>(),
        save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
        total, C, (float)momentum, (float)eps, (int)apply_relu);
    return {y, save_mean, save_rstd};
}

static std::vector<torch::Tensor> bn_nhwc_bwd(
    torch::Tensor x, torch::Tensor dy, torch::Tensor y, torch::Tensor weight,
    torch::Tensor save_mean, torch::Tensor save_rstd,
    bool had_relu, bool want_dres) {
    const int C = x.size(2);
    const long total = x.numel();
    auto dx = torch::empty_like(x);
    auto dres = want_dres ? torch::empty_like(x) : torch::Tensor();
    auto accum = torch::empty({2 * C}, weight.options());
    auto dweight = torch::empty({C}, weight.options());
    auto dbias = torch::empty({C}, weight.options());
    auto stream = at::cuda::getCurrentCUDAStream();
    hipMemsetAsync(accum.data_ptr(), 0, 2 * C * sizeof(float), stream);
    hipLaunchKernelGGL(bn_nhwc_bwd_stats_kernel<short>, dim3(bn_stats_grid(total)), dim3(1024),
        0, stream, (const short*)x.data_ptr(), (const short*)dy.data_ptr(),
        (const short*)y.data_ptr(), accum.data_ptr<float>(),
        save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
        total, C, (int)had_relu);
    hipLaunchKernelGGL(bn_nhwc_bwd_apply_kernel<short>, dim3(bn_grid(total)), dim3(256),
        0, stream, (const short*)x.data_ptr(), (const short*)dy.data_ptr(),
        (const short*)y.data_ptr(), (short*)dx.data_ptr(),
        want_dres ? (short*)dres.data_ptr() : nullptr,
        accum.data_ptr<float>(), weight.data_ptr<float>(),
        save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
        dweight.data_ptr<float>(), dbias.data_ptr<float>(),
        total, C, (int)had_relu);
    return {dx, dweight, dbias, dres};
}

static std::vector<torch::Tensor> target_scan(
    torch::Tensor values, torch::Tensor ret_boot,
    c10::optional<torch::Tensor> rewards, torch::Tensor lambda_,
    c10::optional<torch::Tensor> rhos, c10::optional<torch::Tensor> cs,
    double gamma, int64_t kind)
{
    TORCH_CHECK(values.is_cuda() && values.scalar_type() == torch::kFloat32,
                "target_scan: values must be CUDA float32");
    TORCH_CHECK(values.dim() >= 3, "target_scan: expect (B,T,P,...) layout");
    const int B = values.size(0), T = values.size(1);
    int P = 1;
    for (int d = 2; d < values.dim(); ++d) P *= values.size(d);
    TORCH_CHECK(ret_boot.numel() == (long)B * P,
                "target_scan: bootstrap tensor must have B*P elements");

    auto targets = torch::empty_like(values);
    auto adv = torch::empty_like(values);
    const int n_bp = B * P;
    const int block = 256;
    const int grid = (n_bp + block - 1) / block;
    auto stream = at::cuda::getCurrentCUDAStream();

    const float* rew_ptr = rewards.has_value() ? rewards->data_ptr<float>() : nullptr;
    const float* rho_ptr = rhos.has_value() ? rhos->data_ptr<float>() : nullptr;
    const float* cs_ptr = cs.has_value() ? cs->data_ptr<float>() : nullptr;

    hipLaunchKernelGGL(target_scan_kernel, dim3(grid), dim3(block), 0, stream,
        values.data_ptr<float>(), ret_boot.data_ptr<float>(), rew_ptr,
        lambda_.data_ptr<float>(), rho_ptr, cs_ptr,
        targets.data_ptr<float>(), adv.data_ptr<float>(),
        n_bp, T, P, (float)gamma, (int)kind);
    return {targets, adv};
}

static std::vector<torch::Tensor> masked_sample(
    torch::Tensor logits, torch::Tensor mask, torch::Tensor uniform)
{
    TORCH_CHECK(logits.is_cuda() && logits.dim() == 2, "masked_sample: (N,A) CUDA expected");
    const int N = logits.size(0), A = logits.size(1);
    auto action = torch::empty({N}, logits.options().dtype(torch::kInt64));
    auto prob = torch::empty({N}, logits.options());
    const int block = 256;
    const int grid = (N + block - 1) / block;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(masked_sample_kernel, dim3(grid), dim3(block), 0, stream,
        logits.data_ptr<float>(), mask.data_ptr<float>(), uniform.data_ptr<float>(),
        action.data_ptr<long>(), prob.data_ptr<float>(), N, A);
    return {action, prob};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("target_scan", &target_scan,
          "fused TD/UPGO/V-Trace backward scan (targets, advantages)");
    m.def("masked_sample", &masked_sample,
          "batched masked-softmax action sampling (actions, probs)");
    m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
    m.def("obs_to_nhwc", &obs_to_nhwc, "uint8 NCHW obs -> padded NHWC bf16");
    m.def("obs_to_nhwc_rot", &obs_to_nhwc_rot,
          "canonical uint8 obs -> 4 seat-rotated NHWC bf16 rows per game");
    m.def("torus_conv_fused", &torus_conv_fused,
          "fused wrap-around conv3x3 + BN-fold + residual + relu (MFMA)");
    m.def("torus_wgrad", &torus_wgrad,
          "torus conv weight gradient (MFMA, LDS-staged per image)");
    m.def("loss_head_pre", &loss_head_pre,
          "fused loss pipeline: log-softmax gather + importance ratios + "
          "value/outcome splice");
    m.def("loss_head_fwd", &loss_head_fwd,
          "fused loss pipeline: reduced loss sums (p, v, H, H*decay, dcnt)");
    m.def("loss_head_bwd", &loss_head_bwd,
          "fused loss pipeline backward: d(policy), d(value)");
    m.def("convlstm_cell", &convlstm_cell,
          "fused ConvLSTM cell: implicit-GEMM conv(x|h) + 4-gate state "
          "update, one kernel per cell eval (Geister DRC core)");
    m.def("pack_torus_weights_hip", &pack_torus_weights_hip,
          "pack conv weights into MFMA fragment layout (fwd or dgrad)");
    m.def("bn_nhwc_fwd", &bn_nhwc_fwd,
          "NHWC BN training fwd + fused residual/relu");
    m.def("bn_nhwc_bwd", &bn_nhwc_bwd,
          "NHWC BN training bwd + fused relu mask / residual grad");
    m.def("bn_train_fwd", &bn_train_fwd,
          "BatchNorm2d training forward (y, save_mean, save_rstd)");
    m.def("bn_train_bwd", &bn_train_bwd,
          "BatchNorm2d training backward (dx, dweight, dbias)");
}
