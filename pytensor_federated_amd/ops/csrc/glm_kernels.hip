// CDNA4 (gfx950 / MI355X) fused logp+grad kernels for the federated GLM
// worker models.  Hand-written HIP -- no hipify, no CUDA compatibility.
//
// Replaces the reference's PyTensor-compiled worker functions
// (reference demo_node.py:30-43: pytensor.function([a,b],[logp,*grads]))
// with single-pass, wave64-tiled reductions:
//
//  * fed_gaussian_linear: r = y-(a+b*x); logp, dlogp/da, dlogp/db in ONE
//    pass over x,y (3 reductions fused; HBM-bound, ~4 B/row traffic).
//  * fed_logistic_glm:    z = X.beta row-dot, p = sigmoid(z),
//    logp += y*z - softplus(z), grad += (y-p)*X_row -- X is read EXACTLY
//    once for both logp and grad (the row stays in registers between the
//    dot and the rank-1 grad update).  Per-block grad partials go to a
//    deterministic fp32 slab, reduced by a second tiny kernel.
//
// Numerics: bf16/f32 inputs accumulate fp32 per lane over short spans,
// fp64 across lanes/waves/blocks; f64 inputs accumulate fp64 throughout.
// Outputs are fp64 device buffers shaped for direct RCCL all-reduce:
// [logp, grads...] (the wire layout of common.wrap_logp_grad_func).
//
// Build: hipcc --offload-arch=gfx950 -O3 -shared -fPIC (see ops/build.py).

#include <hip/hip_runtime.h>
#include <math.h>

#define WAVE 64

// ---------------------------------------------------------------------------
// dtype plumbing
// ---------------------------------------------------------------------------

__device__ __forceinline__ float bf16_bits_to_f32(unsigned short u) {
    union { unsigned int i; float f; } cvt;
    cvt.i = ((unsigned int)u) << 16;
    return cvt.f;
}

// 16-byte vector loads: 8 bf16 / 4 f32 / 2 f64 per lane per instruction
// (G13: hipcc does not auto-vectorize bf16 loads; scalar bf16 is ~2x slower).
struct U4 { unsigned int x, y, z, w; };
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_t;

template <typename T> struct VecTraits;
template <> struct VecTraits<float> {
    static constexpr int VEC = 4;
    using acc_t = float;
    __device__ static inline void load(const float* p, float* out) {
        const float4 v = *reinterpret_cast<const float4*>(p);
        out[0] = v.x; out[1] = v.y; out[2] = v.z; out[3] = v.w;
    }
    __device__ static inline float get(const float* p, long long i) { return p[i]; }
    __device__ static inline void unpack(const u32x4_t v, float* out) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            union { unsigned int u; float f; } c;
            c.u = v[j];
            out[j] = c.f;
        }
    }
};
template <> struct VecTraits<double> {
    static constexpr int VEC = 2;
    using acc_t = double;
    __device__ static inline void load(const double* p, double* out) {
        const double2 v = *reinterpret_cast<const double2*>(p);
        out[0] = v.x; out[1] = v.y;
    }
    __device__ static inline double get(const double* p, long long i) { return p[i]; }
};
struct bf16_tag { unsigned short bits; };
template <> struct VecTraits<bf16_tag> {
    static constexpr int VEC = 8;
    using acc_t = float;
    __device__ static inline void load(const bf16_tag* p, float* out) {
        const U4 v = *reinterpret_cast<const U4*>(p);
        const unsigned int w[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            out[2 * j]     = bf16_bits_to_f32((unsigned short)(w[j] & 0xffffu));
            out[2 * j + 1] = bf16_bits_to_f32((unsigned short)(w[j] >> 16));
        }
    }
    __device__ static inline float get(const bf16_tag* p, long long i) {
        return bf16_bits_to_f32(p[i].bits);
    }
    __device__ static inline void unpack(const u32x4_t v, float* out) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            out[2 * j]     = bf16_bits_to_f32((unsigned short)(v[j] & 0xffffu));
            out[2 * j + 1] = bf16_bits_to_f32((unsigned short)(v[j] >> 16));
        }
    }
};

// ---------------------------------------------------------------------------
// cross-lane / cross-wave reduction helpers (wave64!)
// ---------------------------------------------------------------------------

__device__ __forceinline__ double wave_reduce_sum(double v) {
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        v += __shfl_down(v, off, WAVE);
    }
    return v;  // valid in lane 0 of the wave
}

__device__ __forceinline__ float wave_reduce_sum_f32(float v) {
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        v += __shfl_down(v, off, WAVE);
    }
    return v;
}

// Agent-scope write-through (sc1) accessors for cross-workgroup slabs
// (G16 R1's cheap form: sc1 stores + drained ticket on the producer, sc1
// loads on the consumer -- no release/acquire fences, which cost ~1.7 us
// per block and made a fence-based combine 2.7x slower than two kernels).
typedef __attribute__((address_space(1))) unsigned long long gu64_t;
typedef __attribute__((address_space(1))) unsigned gu32_t;

__device__ __forceinline__ void store_sc1_f64(double* p, double v) {
    union { double d; unsigned long long u; } c;
    c.d = v;
    __hip_atomic_store((gu64_t*)p, c.u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ double load_sc1_f64(const double* p) {
    union { unsigned long long u; double d; } c;
    c.u = __hip_atomic_load((const gu64_t*)p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    return c.d;
}

// Block-level fp64 reduction of NACC per-lane values via LDS; the block's
// totals land in lane 0 of wave 0.  BLOCK = 256 threads = 4 waves.
template <int NACC>
__device__ inline void block_reduce_add(double* vals, double* lds /* [4][NACC] */) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
#pragma unroll
    for (int k = 0; k < NACC; ++k) {
        const double w = wave_reduce_sum(vals[k]);
        if (lane == 0) lds[wid * NACC + k] = w;
    }
    __syncthreads();
    if (wid == 0) {
#pragma unroll
        for (int k = 0; k < NACC; ++k) {
            double total = 0.0;
            if (lane == 0) {
#pragma unroll
                for (int w = 0; w < 4; ++w) total += lds[w * NACC + k];
            }
            vals[k] = total;
        }
    }
}

// ---------------------------------------------------------------------------
// Gaussian linear regression: fused logp + d/da + d/db
// ---------------------------------------------------------------------------


// Shared element loop for the gaussian reductions.  Mixed precision: the
// residual r is computed in f64 (an f32 pred carries a systematic rounding
// bias that pollutes the cancellation-heavy sum(r) -- measured ~1e-3
// relative at N=1e7), but the r^2 / r*x PRODUCTS accumulate in f32: their
// rounding is uncorrelated (CLT-cancelling), and per-lane spans are short
// before the f64 cross-lane tree.  Saves ~1/3 of the per-element f64 VALU.
template <typename T>
__device__ __forceinline__ void gauss_accumulate(
    const T* __restrict__ x, const T* __restrict__ y, long long n,
    long long gid, long long gstride, double a, double b,
    double& sr_out, double& srx_out, double& sr2_out
) {
    using TR = VecTraits<T>;
    using A = typename TR::acc_t;
    constexpr int VEC = TR::VEC;
    const long long nvec = n / VEC;
    double sr = 0;
    float srx = 0.f, sr2 = 0.f;
    double srx_d = 0, sr2_d = 0;
    A xv[VEC], yv[VEC];
    long long lane_iters = 0;
    for (long long i = gid; i < nvec; i += gstride) {
        TR::load(x + i * VEC, xv);
        TR::load(y + i * VEC, yv);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
            const double xd = (double)xv[j];
            const double r = (double)yv[j] - (a + b * xd);
            sr += r;
            if constexpr (sizeof(A) == 8) {  // f64 inputs: full f64 products
                srx_d += r * xd;
                sr2_d += r * r;
            } else {
                const float r32 = (float)r;
                srx += r32 * (float)xd;
                sr2 += r32 * r32;
            }
        }
        // bound the f32 partial span (keeps rounding growth ~sqrt(span))
        if (((++lane_iters) & 255) == 0) {
            srx_d += (double)srx;
            sr2_d += (double)sr2;
            srx = 0.f;
            sr2 = 0.f;
        }
    }
    for (long long i = nvec * VEC + gid; i < n; i += gstride) {
        const double xi = (double)TR::get(x, i);
        const double r = (double)TR::get(y, i) - (a + b * xi);
        sr += r;
        if constexpr (sizeof(A) == 8) {
            srx_d += r * xi;
            sr2_d += r * r;
        } else {
            const float r32 = (float)r;
            srx += r32 * (float)xi;
            sr2 += r32 * r32;
        }
    }
    sr_out = sr;
    srx_out = srx_d + (double)srx;
    sr2_out = sr2_d + (double)sr2;
}

// Single-launch variant: pass1 + in-launch combine by the LAST-arriving
// block (agent-scope release/acquire per cdna_hip_programming.md §6 G16;
// saves the finish launch + its ~4.4 us single-block latency).  The ticket
// counter is MONOTONIC: last-of-this-call iff old % gridDim == gridDim-1,
// so no per-call zeroing is needed (grid is a power of two -> the u32 wrap
// stays consistent).  One workspace+ticket per model instance; calls on one
// stream serialize, so slab reuse across calls is safe.
template <typename T>
__global__ __launch_bounds__(256) void k_gaussian_linear_fused(
    const T* __restrict__ x,
    const T* __restrict__ y,
    long long n,
    double a_d,
    double b_d,
    double inv_sig2,
    double logp_const,
    double* __restrict__ slab,       // [gridDim][3]
    unsigned* __restrict__ ticket,   // monotonic arrival counter
    double* __restrict__ out3,       // device result
    double* __restrict__ out3_host,  // mapped pinned mailbox (nullable)
    unsigned long long seq,          // call sequence for the mailbox flag
    const double* __restrict__ theta_dev  // optional [a, b] device buffer
) {
    using TR = VecTraits<T>;
    using A = typename TR::acc_t;
    constexpr int VEC = TR::VEC;
    // theta from device memory enables hipGraph replay with varying theta
    const double a = theta_dev ? theta_dev[0] : a_d;
    const double b = theta_dev ? theta_dev[1] : b_d;

    const long long gid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    const long long gstride = (long long)gridDim.x * blockDim.x;

    double sr, srx, sr2;
    gauss_accumulate<T>(x, y, n, gid, gstride, a, b, sr, srx, sr2);

    __shared__ double lds[4 * 3 + 1];  // ONE shared object (reduce + flag)
    double acc[3] = {sr2, sr, srx};
    block_reduce_add<3>(acc, lds);
    if (threadIdx.x == 0) {
        double* s = slab + 3 * (long long)blockIdx.x;
        // write-through publish: sc1 stores leave L2 immediately; drain this
        // wave's stores before the ticket so the count never overtakes them
        store_sc1_f64(&s[0], acc[0]);
        store_sc1_f64(&s[1], acc[1]);
        store_sc1_f64(&s[2], acc[2]);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        // Sharded arrival tickets (one word saturates at ~88 atomics/us ->
        // 2048 arrivals ~ 23 us measured; 8 group words on separate cache
        // lines + a top word cut that ~8x).  Grouping is by blockIdx
        // parity -- a pure partition, no XCD-placement assumption.
        const unsigned ngroups = gridDim.x < 8 ? gridDim.x : 8;
        const unsigned gsize = gridDim.x / ngroups;
        const unsigned grp = blockIdx.x % ngroups;
        bool last = false;
        const unsigned old = __hip_atomic_fetch_add(
            (gu32_t*)(ticket + 16 * grp), 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        if (old % gsize == gsize - 1) {  // last arrival of this group
            const unsigned t = __hip_atomic_fetch_add(
                (gu32_t*)(ticket + 16 * 8), 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
            last = (t % ngroups) == (ngroups - 1);
        }
        lds[12] = last ? 1.0 : 0.0;
    }
    __syncthreads();
    if (lds[12] == 0.0) return;

    // last-arriving block: reduce every block's slab entry with sc1 loads
    // (L1-bypassing; matches the sc1-stored data, no acquire fence needed)
    double fin[3] = {0.0, 0.0, 0.0};
    for (unsigned i = threadIdx.x; i < gridDim.x; i += blockDim.x) {
#pragma unroll
        for (int k = 0; k < 3; ++k) fin[k] += load_sc1_f64(&slab[3 * (long long)i + k]);
    }
    __syncthreads();  // lds[0..11] reused below
    block_reduce_add<3>(fin, lds);
    if (threadIdx.x == 0) {
        const double v0 = logp_const - 0.5 * inv_sig2 * fin[0];
        const double v1 = inv_sig2 * fin[1];
        const double v2 = inv_sig2 * fin[2];
        out3[0] = v0;
        out3[1] = v1;
        out3[2] = v2;
        if (out3_host != nullptr) {
            out3_host[0] = v0;
            out3_host[1] = v1;
            out3_host[2] = v2;
            __threadfence_system();  // results visible to host before the flag
            ((unsigned long long*)out3_host)[3] = seq;
        }
    }
}

// ---------------------------------------------------------------------------
// Logistic GLM: fused z = X.beta, logp, grad = X^T (y - sigmoid(z))
// ---------------------------------------------------------------------------
//
// One WAVE owns one row at a time (grid-stride over rows).  The row's
// feature chunk assignments: iteration c covers columns
// [c*WAVE*VEC, (c+1)*WAVE*VEC); lane l holds columns c*WAVE*VEC + l*VEC + j.
// For K <= KMAX_REG (held in registers) the row is loaded once and reused
// for the grad update; larger K falls back to a second (L2-hot) read.
//
// Per-block grad partials: each wave accumulates its lanes' column slices
// in registers across all its rows, then waves are summed in LDS and the
// block writes one fp32 slab row: slab[block][K].  k_colsum_reduce sums the
// slab into out[1..K] (deterministic; no fp32 global atomics).

template <typename T, int KITER>  // KITER = K / (WAVE*VEC), compile-time
__global__ __launch_bounds__(256) void k_logistic_glm_reg(
    const T* __restrict__ X,   // [N][K] row-major
    const T* __restrict__ y,
    long long n_rows,
    int K,
    const float* __restrict__ beta,  // [K]
    double* __restrict__ logp_out,   // pre-zeroed scalar
    float* __restrict__ grad_slab    // [gridDim][K]
) {
    using TR = VecTraits<T>;
    constexpr int VEC = TR::VEC;
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int waves_per_block = blockDim.x / WAVE;
    const long long wave_id = (long long)blockIdx.x * waves_per_block + wid;
    const long long n_waves = (long long)gridDim.x * waves_per_block;

    // per-lane register state
    float xreg[KITER][VEC];        // this lane's row slice
    float breg[KITER][VEC];        // this lane's beta slice (loop-invariant)
    float gacc[KITER][VEC];        // this lane's grad accumulator
#pragma unroll
    for (int c = 0; c < KITER; ++c)
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
            breg[c][j] = beta[c * WAVE * VEC + lane * VEC + j];
            gacc[c][j] = 0.f;
        }

    double logp_acc = 0.0;
    // Two rows per iteration: both rows' global loads issue before either
    // row's dependent dot/reduce/transcendental chain, so each wave keeps
    // ~2x the HBM traffic in flight (the 1-row loop measured 68% of HBM
    // peak; the dependent per-row chain was the gap).
    float xreg2[KITER][VEC];
    const long long pair_stride = n_waves * 2;
    long long r = wave_id * 2;
    for (; r + 1 < n_rows; r += pair_stride) {
        const T* row0 = X + r * (long long)K;
        const T* row1 = row0 + K;
        float z0p = 0.f, z1p = 0.f;
#pragma unroll
        for (int c = 0; c < KITER; ++c) {
            // nontemporal: each X row is consumed once (from registers for
            // both logp and grad) -- do not displace L2/L3 lines
            const u32x4_t v0 = __builtin_nontemporal_load(
                (const u32x4_t*)(row0 + c * WAVE * VEC + lane * VEC));
            const u32x4_t v1 = __builtin_nontemporal_load(
                (const u32x4_t*)(row1 + c * WAVE * VEC + lane * VEC));
            TR::unpack(v0, xreg[c]);
            TR::unpack(v1, xreg2[c]);
        }
#pragma unroll
        for (int c = 0; c < KITER; ++c)
#pragma unroll
            for (int j = 0; j < VEC; ++j) {
                z0p += xreg[c][j] * breg[c][j];
                z1p += xreg2[c][j] * breg[c][j];
            }
        float z0 = wave_reduce_sum_f32(z0p);
        float z1 = wave_reduce_sum_f32(z1p);
        z0 = __shfl(z0, 0, WAVE);
        z1 = __shfl(z1, 0, WAVE);
        const float y0 = TR::get(y, r);
        const float y1 = TR::get(y, r + 1);
        // stable: y*z - softplus(z) = y*z - (max(z,0) + log1p(exp(-|z|)))
        const float sp0 = fmaxf(z0, 0.f) + log1pf(__expf(-fabsf(z0)));
        const float sp1 = fmaxf(z1, 0.f) + log1pf(__expf(-fabsf(z1)));
        if (lane == 0) logp_acc += (double)(y0 * z0 - sp0) + (double)(y1 * z1 - sp1);
        const float res0 = y0 - 1.f / (1.f + __expf(-z0));
        const float res1 = y1 - 1.f / (1.f + __expf(-z1));
#pragma unroll
        for (int c = 0; c < KITER; ++c)
#pragma unroll
            for (int j = 0; j < VEC; ++j)
                gacc[c][j] += res0 * xreg[c][j] + res1 * xreg2[c][j];
    }
    for (; r < n_rows; r += pair_stride) {  // odd tail row of this wave
        const T* row = X + r * (long long)K;
        float z_part = 0.f;
#pragma unroll
        for (int c = 0; c < KITER; ++c) {
            TR::load(row + c * WAVE * VEC + lane * VEC, xreg[c]);
#pragma unroll
            for (int j = 0; j < VEC; ++j) z_part += xreg[c][j] * breg[c][j];
        }
        float z = wave_reduce_sum_f32(z_part);
        z = __shfl(z, 0, WAVE);
        const float yr = TR::get(y, r);
        const float sp = fmaxf(z, 0.f) + log1pf(__expf(-fabsf(z)));
        if (lane == 0) logp_acc += (double)(yr * z - sp);
        const float resid = yr - 1.f / (1.f + __expf(-z));
#pragma unroll
        for (int c = 0; c < KITER; ++c)
#pragma unroll
            for (int j = 0; j < VEC; ++j) gacc[c][j] += resid * xreg[c][j];
    }

    // cross-wave grad reduction in LDS (dynamic: K floats), then one slab row
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* g_lds = reinterpret_cast<float*>(smem);                  // [K]
    double* l_lds = reinterpret_cast<double*>(smem + ((K * 4 + 15) & ~15));  // [waves]
    for (int w = 0; w < waves_per_block; ++w) {
        if (wid == w) {
#pragma unroll
            for (int c = 0; c < KITER; ++c)
#pragma unroll
                for (int j = 0; j < VEC; ++j) {
                    const int col = c * WAVE * VEC + lane * VEC + j;
                    if (w == 0)
                        g_lds[col] = gacc[c][j];
                    else
                        g_lds[col] += gacc[c][j];
                }
        }
        __syncthreads();
    }
    float* slab_row = grad_slab + (long long)blockIdx.x * K;
    for (int col = threadIdx.x; col < K; col += blockDim.x) slab_row[col] = g_lds[col];

    if (lane == 0) l_lds[wid] = logp_acc;
    __syncthreads();
    if (threadIdx.x == 0) {
        double t = 0.0;
        for (int w = 0; w < waves_per_block; ++w) t += l_lds[w];
        atomicAdd(logp_out, t);
    }
}

// Column-wise fp64 reduction of the grad slab into out[1..K].
// grid = (ceil(K/256), CHUNKS): blockIdx.y sums a slice of the slab rows and
// fp64-atomicAdds its column partials (CHUNKS atomics per address -- cheap;
// a 1-d grid is latency-bound: 4 blocks reading 8 MB measured 363 us).
__global__ __launch_bounds__(256) void k_colsum_reduce(
    const float* __restrict__ slab,  // [n_slabs][K]
    int n_slabs,
    int K,
    double* __restrict__ out_grad  // [K], pre-zeroed
) {
    const int col = blockIdx.x * blockDim.x + threadIdx.x;
    if (col >= K) return;
    const int chunk = (n_slabs + gridDim.y - 1) / gridDim.y;
    const int b0 = blockIdx.y * chunk;
    const int b1 = min(b0 + chunk, n_slabs);
    double s = 0.0;
    for (int b = b0; b < b1; ++b) s += (double)slab[(long long)b * K + col];
    if (gridDim.y == 1)
        out_grad[col] = s;
    else
        atomicAdd(&out_grad[col], s);
}

// ---------------------------------------------------------------------------
// C entry points (ctypes-friendly; orchestration incl. zeroing + launches)
// ---------------------------------------------------------------------------

enum FedDtype { FED_F32 = 0, FED_F64 = 1, FED_BF16 = 2 };

#include <stdlib.h>

static int grid_cap(void) {
    static int cap = 0;
    if (cap == 0) {
        const char* env = getenv("FED_GRID_CAP");
        cap = env ? atoi(env) : 1024;  // G11: cap + grid-stride (A/B on MI355X: 512-1024 best)
        if (cap < 1 || cap > 8192) cap = 2048;
    }
    return cap;
}

static inline int pick_grid(long long work_items, int block) {
    long long blocks = (work_items + block - 1) / block;
    if (blocks > grid_cap()) blocks = grid_cap();
    if (blocks < 1) blocks = 1;
    return (int)blocks;
}

extern "C" {

const char* fed_last_hip_error(void) { return hipGetErrorString(hipGetLastError()); }

// workspace layout (fp64 words): [0..72) = 9 ticket words, 64 B apart
// (8 groups + 1 top; MUST be zero-initialized once), [72..72+3*grid) =
// fp64 partial slab.  ws_bytes >= (72 + 3*grid_max) * 8.
static int gaussian_linear_impl(
    const void* x, const void* y, long long n,
    double a, double b, double sigma,
    double* out3, double* out3_host,
    double* workspace, long long ws_bytes,
    int dtype, hipStream_t stream, unsigned long long seq,
    const double* theta_dev = nullptr
) {
    const double inv_sig2 = 1.0 / (sigma * sigma);
    const double logp_const = -0.5 * (double)n * log(2.0 * M_PI * sigma * sigma);
    const int block = 256;
    int grid;
    switch (dtype) {
        case FED_F32: grid = pick_grid(n / 4, block); break;
        case FED_F64: grid = pick_grid(n / 2, block); break;
        case FED_BF16: grid = pick_grid(n / 8, block); break;
        default: return -2;
    }
    long long ws_cap = (ws_bytes / 8 - 72) / 3;
    if (grid > ws_cap) grid = (int)ws_cap;
    if (grid < 1) return -3;
    // power of two so the monotonic u32 tickets wrap consistently
    while (grid & (grid - 1)) grid &= grid - 1;
    unsigned* ticket = (unsigned*)workspace;  // 9 words, 64 B apart
    double* slab = workspace + 72;
    switch (dtype) {
        case FED_F32:
            hipLaunchKernelGGL(k_gaussian_linear_fused<float>, dim3(grid), dim3(block), 0, stream,
                               (const float*)x, (const float*)y, n, a, b, inv_sig2, logp_const,
                               slab, ticket, out3, out3_host, seq, theta_dev);
            break;
        case FED_F64:
            hipLaunchKernelGGL(k_gaussian_linear_fused<double>, dim3(grid), dim3(block), 0, stream,
                               (const double*)x, (const double*)y, n, a, b, inv_sig2, logp_const,
                               slab, ticket, out3, out3_host, seq, theta_dev);
            break;
        case FED_BF16:
            hipLaunchKernelGGL(k_gaussian_linear_fused<bf16_tag>, dim3(grid), dim3(block), 0, stream,
                               (const bf16_tag*)x, (const bf16_tag*)y, n, a, b, inv_sig2, logp_const,
                               slab, ticket, out3, out3_host, seq, theta_dev);
            break;
    }
    return (int)hipGetLastError();
}

int fed_gaussian_linear(
    const void* x, const void* y, long long n,
    double a, double b, double sigma,
    double* out3, double* workspace, long long ws_bytes,
    int dtype, void* stream_v
) {
    return gaussian_linear_impl(x, y, n, a, b, sigma, out3, nullptr,
                                workspace, ws_bytes, dtype, (hipStream_t)stream_v, 0);
}

// Synchronous single-call evaluation: ONE launch; the last-arriving block
// writes {logp, ga, gb, seq} into the mapped pinned mailbox; the host
// spin-reads the seq flag (no hipStreamSynchronize on the happy path).
int fed_gaussian_linear_eval(
    const void* x, const void* y, long long n,
    double a, double b, double sigma,
    double* out3_dev, double* out3_host,
    double* workspace, long long ws_bytes,
    int dtype, void* stream_v, unsigned long long seq
) {
    hipStream_t stream = (hipStream_t)stream_v;
    void* mailbox_dev = nullptr;  // device-side alias of the pinned mailbox
    hipError_t perr = hipHostGetDevicePointer(&mailbox_dev, out3_host, 0);
    if (perr != hipSuccess) return (int)perr;
    int rc = gaussian_linear_impl(x, y, n, a, b, sigma, out3_dev,
                                  (double*)mailbox_dev, workspace, ws_bytes,
                                  dtype, stream, seq);
    if (rc != 0) return rc;
    volatile unsigned long long* flag = ((volatile unsigned long long*)out3_host) + 3;
    for (long long spins = 0; spins < 400000000LL; ++spins) {  // ~>1 s bound
        if (*flag == seq) return 0;
    }
    // flag never arrived: drain the stream and surface the real error
    hipError_t serr = hipStreamSynchronize(stream);
    if (serr != hipSuccess) return (int)serr;
    return (*flag == seq) ? 0 : -5;
}

// Mapped pinned host memory for the GPU-written result mailbox.
void* fed_host_alloc(long long bytes) {
    void* p = nullptr;
    if (hipHostMalloc(&p, bytes, hipHostMallocMapped) != hipSuccess) return nullptr;
    return p;
}

int fed_host_get_device_ptr(void* host_ptr, void** dev_ptr) {
    return (int)hipHostGetDevicePointer(dev_ptr, host_ptr, 0);
}

// out = fp64[1+K] = {logp, grad...}; beta_f32 = fp32[K] device;
// workspace holds the fp32 grad slab (ws_bytes >= grid*K*4).
int fed_logistic_glm(
    const void* X, const void* y, long long n_rows, int K,
    const float* beta_f32,
    double* out, float* workspace, long long ws_bytes,
    int dtype, void* stream_v
) {
    hipStream_t stream = (hipStream_t)stream_v;
    hipError_t err = hipMemsetAsync(out, 0, (1 + K) * sizeof(double), stream);
    if (err != hipSuccess) return (int)err;
    const int block = 256;
    const int waves_per_block = block / WAVE;
    int grid = pick_grid(n_rows / waves_per_block + 1, 1);
    if (grid > 1024) grid = 1024;
    const long long need = (long long)grid * K * sizeof(float);
    if (need > ws_bytes) grid = (int)(ws_bytes / ((long long)K * sizeof(float)));
    if (grid < 1) return -3;
    const int lds_bytes = ((K * 4 + 15) & ~15) + waves_per_block * 8;

#define LAUNCH_LOGISTIC(T, KITER)                                                      \
    hipLaunchKernelGGL((k_logistic_glm_reg<T, KITER>), dim3(grid), dim3(block),        \
                       lds_bytes, stream, (const T*)X, (const T*)y, n_rows, K,         \
                       beta_f32, out, workspace)

    if (dtype == FED_BF16) {
        switch (K) {
            case 512:  LAUNCH_LOGISTIC(bf16_tag, 1); break;
            case 1024: LAUNCH_LOGISTIC(bf16_tag, 2); break;
            case 2048: LAUNCH_LOGISTIC(bf16_tag, 4); break;
            default: return -4;  // K must be a multiple of 512 (wave64 x bf16x8)
        }
    } else if (dtype == FED_F32) {
        switch (K) {
            case 256:  LAUNCH_LOGISTIC(float, 1); break;
            case 512:  LAUNCH_LOGISTIC(float, 2); break;
            case 1024: LAUNCH_LOGISTIC(float, 4); break;
            default: return -4;
        }
    } else {
        return -2;
    }
#undef LAUNCH_LOGISTIC
    hipError_t kerr = hipGetLastError();
    if (kerr != hipSuccess) return (int)kerr;

    const int rgrid = (K + 255) / 256;
    int chunks = grid / 8;
    if (chunks < 1) chunks = 1;
    if (chunks > 64) chunks = 64;
    hipLaunchKernelGGL(k_colsum_reduce, dim3(rgrid, chunks), dim3(256), 0, stream,
                       workspace, grid, K, out + 1);
    return (int)hipGetLastError();
}

}  // extern "C"


// Publish a small fp64 result vector to the pinned mailbox with a
// self-incrementing epoch flag -- replayable from a hipGraph (the epoch
// lives in device memory, not in baked kernel args).
__global__ __launch_bounds__(64) void k_publish_result(
    const double* __restrict__ buf, int n,
    double* __restrict__ mailbox,            // [n results][1 seq slot]
    unsigned long long* __restrict__ epoch
) {
    if (threadIdx.x == 0) {
        for (int k = 0; k < n; ++k) mailbox[k] = buf[k];
        const unsigned long long e = *epoch + 1;
        *epoch = e;
        __threadfence_system();
        ((unsigned long long*)mailbox)[n] = e;
    }
}

extern "C" int fed_publish_result(
    const double* buf, int n, double* mailbox_host,
    unsigned long long* epoch_dev, void* stream_v
) {
    void* mailbox_dev = nullptr;
    hipError_t perr = hipHostGetDevicePointer(&mailbox_dev, mailbox_host, 0);
    if (perr != hipSuccess) return (int)perr;
    hipLaunchKernelGGL(k_publish_result, dim3(1), dim3(64), 0,
                       (hipStream_t)stream_v, buf, n, (double*)mailbox_dev, epoch_dev);
    return (int)hipGetLastError();
}

// ---------------------------------------------------------------------------
// Native multi-shard linear engine: N shards on one GPU, one call per eval
// ---------------------------------------------------------------------------
//
// The native successor of the reference's ParallelAsyncOp fan-out
// (reference op_async.py:107-132): N federated shards resident on one
// MI355X evaluate CONCURRENTLY, each on its own HIP stream; a combine
// kernel (gated by per-shard events) sums the per-shard [logp, ga, gb]
// and writes the total to the device buffer + pinned mailbox.  The whole
// fan-out costs ONE host call; the host polls the mailbox seq flag.

__global__ __launch_bounds__(64) void k_sum_shards3(
    const double* __restrict__ shard_out,  // [n_shards][3]
    int n_shards,
    double* __restrict__ out3,
    double* __restrict__ out3_host,
    unsigned long long seq
) {
    if (threadIdx.x == 0) {
        double v[3] = {0.0, 0.0, 0.0};
        for (int s = 0; s < n_shards; ++s)
            for (int k = 0; k < 3; ++k) v[k] += shard_out[3 * s + k];
        out3[0] = v[0]; out3[1] = v[1]; out3[2] = v[2];
        if (out3_host != nullptr) {
            out3_host[0] = v[0]; out3_host[1] = v[1]; out3_host[2] = v[2];
            __threadfence_system();
            ((unsigned long long*)out3_host)[3] = seq;
        }
    }
}

#define FED_MAX_SHARDS 64

struct FedLinearEngine {
    int n_shards;
    int dtype;
    double sigma;
    const void* xs[FED_MAX_SHARDS];
    const void* ys[FED_MAX_SHARDS];
    long long ns[FED_MAX_SHARDS];
    hipStream_t streams[FED_MAX_SHARDS];
    hipEvent_t events[FED_MAX_SHARDS];
    double* ws[FED_MAX_SHARDS];      // per-shard ticket+slab workspace
    double* shard_out;               // [n_shards][3] device
    double* out3;                    // device total
    double* mailbox;                 // pinned mapped {3 results, seq}
    double* mailbox_dev;
    unsigned long long seq;
};

extern "C" {

int fed_linear_engine_destroy(void* handle);

// hipGraph-capturable launch: theta read from a device buffer at kernel
// execution time (so a captured graph replays with updated theta).
int fed_gaussian_linear_theta(
    const void* x, const void* y, long long n,
    const double* theta_dev, double sigma,
    double* out3, double* workspace, long long ws_bytes,
    int dtype, void* stream_v
) {
    return gaussian_linear_impl(x, y, n, 0.0, 0.0, sigma, out3, nullptr,
                                workspace, ws_bytes, dtype, (hipStream_t)stream_v,
                                0, theta_dev);
}

void* fed_linear_engine_create(
    int n_shards, const void** xs, const void** ys, const long long* ns,
    double sigma, int dtype
) {
    if (n_shards < 1 || n_shards > FED_MAX_SHARDS) return nullptr;
    FedLinearEngine* e = new FedLinearEngine();
    e->n_shards = n_shards;
    e->dtype = dtype;
    e->sigma = sigma;
    e->seq = 0;
    const long long ws_words = 72 + 3 * 2048;
    for (int s = 0; s < n_shards; ++s) {
        e->xs[s] = xs[s];
        e->ys[s] = ys[s];
        e->ns[s] = ns[s];
        if (hipStreamCreateWithFlags(&e->streams[s], hipStreamNonBlocking) != hipSuccess ||
            hipEventCreateWithFlags(&e->events[s], hipEventDisableTiming) != hipSuccess ||
            hipMalloc(&e->ws[s], ws_words * 8) != hipSuccess ||
            hipMemset(e->ws[s], 0, ws_words * 8) != hipSuccess) {
            fed_linear_engine_destroy(e);
            return nullptr;
        }
    }
    if (hipMalloc(&e->shard_out, n_shards * 3 * 8) != hipSuccess ||
        hipMalloc(&e->out3, 3 * 8) != hipSuccess ||
        hipHostMalloc((void**)&e->mailbox, 4 * 8, hipHostMallocMapped) != hipSuccess ||
        hipHostGetDevicePointer((void**)&e->mailbox_dev, e->mailbox, 0) != hipSuccess) {
        fed_linear_engine_destroy(e);
        return nullptr;
    }
    e->mailbox[3] = 0.0;
    return e;
}

// One federated fan-out evaluation; blocks until the summed [logp, ga, gb]
// is in out3_host_result.  sync_stream (torch current stream) is made to
// wait for the combine so subsequent torch work stays ordered.
int fed_linear_engine_eval(
    void* handle, double a, double b, double* out3_host_result, void* sync_stream
) {
    FedLinearEngine* e = (FedLinearEngine*)handle;
    e->seq += 1;
    for (int s = 0; s < e->n_shards; ++s) {
        int rc = gaussian_linear_impl(
            e->xs[s], e->ys[s], e->ns[s], a, b, e->sigma,
            e->shard_out + 3 * s, nullptr,
            e->ws[s], (72 + 3 * 2048) * 8, e->dtype, e->streams[s], 0);
        if (rc != 0) return rc;
        if (hipEventRecord(e->events[s], e->streams[s]) != hipSuccess) return -7;
    }
    for (int s = 1; s < e->n_shards; ++s)
        if (hipStreamWaitEvent(e->streams[0], e->events[s], 0) != hipSuccess) return -8;
    hipLaunchKernelGGL(k_sum_shards3, dim3(1), dim3(64), 0, e->streams[0],
                       e->shard_out, e->n_shards, e->out3, e->mailbox_dev, e->seq);
    hipError_t kerr = hipGetLastError();
    if (kerr != hipSuccess) return (int)kerr;
    if (sync_stream != nullptr) {
        if (hipEventRecord(e->events[0], e->streams[0]) != hipSuccess) return -7;
        if (hipStreamWaitEvent((hipStream_t)sync_stream, e->events[0], 0) != hipSuccess)
            return -8;
    }
    volatile unsigned long long* flag = ((volatile unsigned long long*)e->mailbox) + 3;
    for (long long spins = 0; spins < 400000000LL; ++spins) {
        if (*flag == e->seq) {
            out3_host_result[0] = e->mailbox[0];
            out3_host_result[1] = e->mailbox[1];
            out3_host_result[2] = e->mailbox[2];
            return 0;
        }
    }
    hipError_t serr = hipStreamSynchronize(e->streams[0]);
    if (serr != hipSuccess) return (int)serr;
    if (*flag != e->seq) return -5;
    out3_host_result[0] = e->mailbox[0];
    out3_host_result[1] = e->mailbox[1];
    out3_host_result[2] = e->mailbox[2];
    return 0;
}

int fed_linear_engine_destroy(void* handle) {
    FedLinearEngine* e = (FedLinearEngine*)handle;
    for (int s = 0; s < e->n_shards; ++s) {
        if (e->ws[s]) (void)hipFree(e->ws[s]);
        if (e->streams[s]) (void)hipStreamDestroy(e->streams[s]);
        if (e->events[s]) (void)hipEventDestroy(e->events[s]);
    }
    if (e->shard_out) (void)hipFree(e->shard_out);
    if (e->out3) (void)hipFree(e->out3);
    if (e->mailbox) (void)hipHostFree(e->mailbox);
    delete e;
    return 0;
}

}  // extern "C"

// ---------------------------------------------------------------------------
// Device-buffer IPC plumbing for the on-node device codec (npproto.device)
// ---------------------------------------------------------------------------
// Replaces the host-copy wire path for same-node processes: arrays move
// HBM->HBM through dmabuf IPC handles (HSA_ENABLE_IPC_MODE_LEGACY=0)
// instead of D2H -> protobuf -> H2D.  ("HIP device-buffer
// codec ... device pointer or IPC handle", replacing npproto/utils.py:9-24's
// bytes() host copy.)

extern "C" {

void* fed_device_alloc(long long bytes) {
    void* p = nullptr;
    if (hipMalloc(&p, bytes) != hipSuccess) return nullptr;
    return p;
}

int fed_device_free(void* p) { return (int)hipFree(p); }

int fed_d2d_copy(void* dst, const void* src, long long bytes, void* stream_v) {
    return (int)hipMemcpyAsync(dst, src, bytes, hipMemcpyDeviceToDevice,
                               (hipStream_t)stream_v);
}

int fed_h2d_copy(void* dst, const void* src, long long bytes, void* stream_v) {
    return (int)hipMemcpyAsync(dst, src, bytes, hipMemcpyHostToDevice,
                               (hipStream_t)stream_v);
}

int fed_stream_sync(void* stream_v) {
    return (int)hipStreamSynchronize((hipStream_t)stream_v);
}

int fed_ipc_get_handle(void* dev_ptr, unsigned char* handle64) {
    hipIpcMemHandle_t h;
    hipError_t err = hipIpcGetMemHandle(&h, dev_ptr);
    if (err != hipSuccess) return (int)err;
    __builtin_memcpy(handle64, &h, sizeof(h));
    return 0;
}

int fed_ipc_open(const unsigned char* handle64, void** dev_ptr) {
    hipIpcMemHandle_t h;
    __builtin_memcpy(&h, handle64, sizeof(h));
    return (int)hipIpcOpenMemHandle(dev_ptr, h, hipIpcMemLazyEnablePeerAccess);
}

int fed_ipc_close(void* dev_ptr) { return (int)hipIpcCloseMemHandle(dev_ptr); }

}  // extern "C"

// ---------------------------------------------------------------------------
// MFMA fragment-map probe (test-only): D = A[16x32] . B[32x16], bf16->f32.
// Assumed lane mapping (verified on hardware by tests/test_gpu.py):
//   A: row i = lane&15, k = (lane>>4)*8 + j   (8 bf16 per lane)
//   B: col j = lane&15, k = (lane>>4)*8 + jj  (8 bf16 per lane)
//   D: col   = lane&15, row = (lane>>4)*4 + reg
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ __launch_bounds__(64) void k_mfma_probe(
    const unsigned short* __restrict__ A,  // [16][32] bf16 row-major
    const unsigned short* __restrict__ B,  // [32][16] bf16 row-major
    float* __restrict__ D                  // [16][16] f32 row-major
) {
    const int lane = threadIdx.x & 63;
    union { bf16x8 v; unsigned short u[8]; } a_frag, b_frag;
    const int arow = lane & 15;
    const int k0 = (lane >> 4) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        a_frag.u[j] = A[arow * 32 + k0 + j];
        b_frag.u[j] = B[(k0 + j) * 16 + (lane & 15)];
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag.v, b_frag.v, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int row = (lane >> 4) * 4 + r;
        const int col = lane & 15;
        D[row * 16 + col] = acc[r];
    }
}

extern "C" int fed_mfma_probe(const void* A, const void* B, void* D, void* stream_v) {
    hipLaunchKernelGGL(k_mfma_probe, dim3(1), dim3(64), 0, (hipStream_t)stream_v,
                       (const unsigned short*)A, (const unsigned short*)B, (float*)D);
    return (int)hipGetLastError();
}

// ---------------------------------------------------------------------------
// Batched logistic GLM: B=16 chains per call on MFMA matrix cores
// ---------------------------------------------------------------------------
//
// Multi-chain MCMC (the reference's pm.sample(cores=N) axis) evaluates 16
// proposal vectors in ONE pass over X:
//
//     Z[rows,16]  = X[rows,K] . Theta[K,16]        (phase A, MFMA)
//     logp[b]    += sum_r  y z - softplus(z)
//     R[rows,16]  = y - sigmoid(Z)
//     G[K,16]    += X^T[K,rows] . R[rows,16]       (phase B, MFMA)
//
// X is HBM-read once per call (k-chunks staged to LDS, phase B re-reads the
// L2-hot chunk); at B=16 the arithmetic is 4*N*K*B flops -- VALU could not
// keep up with the HBM stream (0.8 TFLOP/call vs ~4 ms of traffic), MFMA
// makes compute a ~10% bystander.  Per-chain cost is ~B x lower than the
// single-chain kernel.
//
// Geometry: block = 256 threads = 4 waves; row tile 64 (16 rows/wave in
// phase A); K chunked by 128 (each wave owns a 32-col slice in phase B);
// G accumulates in AGPRs (16 tiles x 4 f32/lane/wave); per-block partials
// go to an fp32 slab [16 logp | K*16 G], reduced by k_colsum_reduce.
//
// Verified fragment maps (fed_mfma_probe): A[i=l&15][k=(l>>4)*8+j],
// B[k=(l>>4)*8+j][j=l&15], D[col=l&15][row=(l>>4)*4+r].

#define BCH 16           // chains per call
#define BL_ROWS 64       // rows per block tile
#define BL_CHUNK 128     // K columns per staged chunk
#define XPAD 8           // X_lds row pad (elems): stride 272 B, b128-clean
#define TPAD 8           // Theta row pad: stride 2064+16 B
#define RPAD 8           // R_T row pad: stride 144 B

template <int K>  // compile-time K: keeps g_acc statically indexed (no
                  // scratch spill -- cdna_hip_programming.md rule 20)
__global__ __launch_bounds__(256) void k_logistic_glm_batched(
    const unsigned short* __restrict__ X,   // [N][K] bf16
    const unsigned short* __restrict__ y,   // [N] bf16
    long long n_rows,
    const unsigned short* __restrict__ theta_t,  // [BCH][K] bf16 (transposed!)
    float* __restrict__ slab                     // [grid][BCH + K*BCH]
) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    constexpr int n_chunks = K / BL_CHUNK;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned short* th_lds = (unsigned short*)smem;                 // [BCH][K+TPAD]
    const int th_stride = K + TPAD;
    unsigned short* x_lds = th_lds + BCH * th_stride;               // [2][BL_ROWS][BL_CHUNK+XPAD]
    const int x_stride = BL_CHUNK + XPAD;
    const int x_buf = BL_ROWS * x_stride;
    unsigned short* rt_lds = x_lds + 2 * x_buf;                     // [BCH][BL_ROWS+RPAD]
    const int rt_stride = BL_ROWS + RPAD;
    float* y_lds = (float*)(rt_lds + BCH * rt_stride + 8);          // [BL_ROWS]
    float* red_lds = y_lds + BL_ROWS;                               // [256]

    // ---- stage Theta^T once per block ----
    for (int idx = threadIdx.x * 8; idx < BCH * K; idx += 256 * 8) {
        const int b = idx / K;
        const int k = idx % K;
        *(U4*)&th_lds[b * th_stride + k] = *(const U4*)&theta_t[b * K + k];
    }

    typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
    typedef __attribute__((ext_vector_type(4))) float f32x4_t;
    union frag_u { bf16x8_t v; unsigned short u[8]; U4 q; };

    f32x4_t g_acc[n_chunks * 2];
#pragma unroll
    for (int t = 0; t < n_chunks * 2; ++t) g_acc[t] = (f32x4_t){0.f, 0.f, 0.f, 0.f};
    float logp_acc = 0.f;  // this lane's chain partial (chain = lane&15)

    // staging helpers: thread owns rows {r0, r0+16, r0+32, r0+48}, 16 B each
    const int st_r0 = threadIdx.x / 16;          // 0..15
    const int st_k0 = (threadIdx.x % 16) * (BL_CHUNK / 16);

    const long long n_tiles = (n_rows + BL_ROWS - 1) / BL_ROWS;
    for (long long tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
        const long long row0 = tile * BL_ROWS;
        if (threadIdx.x < BL_ROWS) {
            const long long r = row0 + threadIdx.x;
            y_lds[threadIdx.x] = r < n_rows ? bf16_bits_to_f32(y[r]) : 0.f;
        }

        // double-buffered chunk pipeline over BOTH phases: phase A consumes
        // chunks 0..n-1 (Z), phase B consumes them again (G).  One barrier
        // per step; next chunk's global loads issue before the MFMAs.
        U4 ld[4];
#define LOAD_CHUNK(c)                                                              _Pragma("unroll") for (int rr = 0; rr < 4; ++rr) {                                 const long long row = row0 + st_r0 + rr * 16;                                  ld[rr] = (U4){0, 0, 0, 0};                                                     if (row < n_rows)                                                                  ld[rr] = *(const U4*)&X[row * (long long)K + (c) * BL_CHUNK + st_k0];     }
#define WRITE_CHUNK(buf)                                                           _Pragma("unroll") for (int rr = 0; rr < 4; ++rr)                                   *(U4*)&x_lds[(buf) * x_buf + (st_r0 + rr * 16) * x_stride + st_k0] = ld[rr];

        LOAD_CHUNK(0)
        WRITE_CHUNK(0)
        int cur = 0;

        // ---- phase A: Z = X . Theta ----
        f32x4_t z_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int c = 0; c < n_chunks; ++c) {
            __syncthreads();  // buf[cur] (and Theta on c==0) visible
            if (c + 1 < n_chunks) LOAD_CHUNK(c + 1)
#pragma unroll
            for (int ks = 0; ks < BL_CHUNK / 32; ++ks) {
                frag_u a, b;
                const int arow = wid * 16 + (lane & 15);
                const int ak = ks * 32 + (lane >> 4) * 8;
                a.q = *(U4*)&x_lds[cur * x_buf + arow * x_stride + ak];
                const int bk = c * BL_CHUNK + ks * 32 + (lane >> 4) * 8;
                b.q = *(U4*)&th_lds[(lane & 15) * th_stride + bk];
                z_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, z_acc, 0, 0, 0);
            }
            if (c + 1 < n_chunks) WRITE_CHUNK(cur ^ 1)
            cur ^= 1;
        }

        // ---- logp + R from Z (also prefetch chunk 0 for phase B) ----
        LOAD_CHUNK(0)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int row_in_wave = (lane >> 4) * 4 + r;
            const int row_in_tile = wid * 16 + row_in_wave;
            const long long row = row0 + row_in_tile;
            const int chain = lane & 15;
            float z = z_acc[r];
            float yv = y_lds[row_in_tile];
            float resid = 0.f;
            if (row < n_rows) {
                const float sp = fmaxf(z, 0.f) + log1pf(__expf(-fabsf(z)));
                logp_acc += yv * z - sp;
                resid = yv - 1.f / (1.f + __expf(-z));
            }
            union { float f; unsigned int u; } cv;
            cv.f = resid;
            const unsigned int rnd = 0x7fff + ((cv.u >> 16) & 1);
            rt_lds[chain * rt_stride + row_in_tile] = (unsigned short)((cv.u + rnd) >> 16);
        }
        __syncthreads();  // R complete; x_lds free
        WRITE_CHUNK(0)
        cur = 0;

        // ---- phase B: G += X_chunk^T . R (chunks are L2-hot) ----
#pragma unroll
        for (int c = 0; c < n_chunks; ++c) {
            __syncthreads();
            if (c + 1 < n_chunks) LOAD_CHUNK(c + 1)
#pragma unroll
            for (int t2 = 0; t2 < 2; ++t2) {
                const int kcol0 = wid * 32 + t2 * 16;
                f32x4_t acc = g_acc[c * 2 + t2];
#pragma unroll
                for (int rs = 0; rs < 2; ++rs) {
                    frag_u a, b;
                    const int kcol = kcol0 + (lane & 15);
                    const int arow0 = rs * 32 + (lane >> 4) * 8;
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        a.u[j] = x_lds[cur * x_buf + (arow0 + j) * x_stride + kcol];
                    b.q = *(U4*)&rt_lds[(lane & 15) * rt_stride + rs * 32 + (lane >> 4) * 8];
                    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
                }
                g_acc[c * 2 + t2] = acc;
            }
            if (c + 1 < n_chunks) WRITE_CHUNK(cur ^ 1)
            cur ^= 1;
        }
        __syncthreads();  // rt_lds reuse next tile
#undef LOAD_CHUNK
#undef WRITE_CHUNK
    }

    // ---- epilogue: block partials -> slab ----
    red_lds[threadIdx.x] = logp_acc;
    __syncthreads();
    float* slab_blk = slab + (long long)blockIdx.x * (BCH + (long long)K * BCH);
    if (threadIdx.x < BCH) {
        float s = 0.f;
        for (int i = threadIdx.x; i < 256; i += BCH) s += red_lds[i];
        slab_blk[threadIdx.x] = s;
    }
    float* g_slab = slab_blk + BCH;
#pragma unroll
    for (int c = 0; c < n_chunks; ++c) {
#pragma unroll
        for (int t2 = 0; t2 < 2; ++t2) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int kcol = c * BL_CHUNK + wid * 32 + t2 * 16 + (lane >> 4) * 4 + r;
                const int chain = lane & 15;
                g_slab[(long long)kcol * BCH + chain] = g_acc[c * 2 + t2][r];
            }
        }
    }
}


// ---------------------------------------------------------------------------
// Batched logistic v2: fewer barriers, swizzled LDS, write-after-barrier
// ---------------------------------------------------------------------------
// Round-1 PMC on v1 (profiles/PROFILES.md): 60% wave-park cycles (staging
// written BEFORE each chunk barrier, loads only get the MFMA span) and
// 10.8% LDS bank-conflict cycles (phase-B transposed u16 reads: rows r/r+8
// land on one bank at the 16B-aligned row stride).  v2 changes:
//
//  * (a write-after-barrier staging variant was tried and REVERTED: hipcc
//    coalesced the 4 staging loads into one register quad with a serial
//    vmcnt(0) per load -- 2.8 ms vs 1.57.  v1's load placement, chunk
//    c+1's loads issued before chunk c's MFMAs, is what keeps 4 loads in
//    flight under this compiler.)
//  * XOR swizzle of the 16-byte group index, key = ((row>>3)&1)<<1 --
//    rows r and r+8 (the colliding pair within a 32-lane half) get their
//    group-bit-1 flipped against each other, separating their banks for
//    the phase-B scalar reads while leaving the b128 write/read patterns
//    near-conflict-free.
//  * edge rows keep v1's conditional zero-fill: an address-clamp variant
//    (branch-free) made hipcc spill the staged loads to scratch with a
//    vmcnt(0) after EACH -- serial HBM round trips (2.8 ms).  With the
//    conditional form the 4 loads stay in 4 register quads, in flight
//    together (verified in the ISA).

template <int K>
__global__ __launch_bounds__(256) void k_logistic_glm_batched_v2(
    const unsigned short* __restrict__ X,   // [N][K] bf16
    const unsigned short* __restrict__ y,   // [N] bf16
    long long n_rows,
    const unsigned short* __restrict__ theta_t,  // [BCH][K] bf16 (transposed)
    float* __restrict__ slab                     // [grid][BCH + K*BCH]
) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    constexpr int n_chunks = K / BL_CHUNK;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned short* th_lds = (unsigned short*)smem;             // [BCH][K+TPAD]
    const int th_stride = K + TPAD;
    unsigned short* x_lds = th_lds + BCH * th_stride;           // [2][BL_ROWS][BL_CHUNK+XPAD]
    const int x_stride = BL_CHUNK + XPAD;
    const int x_buf = BL_ROWS * x_stride;
    unsigned short* rt_lds = x_lds + 2 * x_buf;                 // [BCH][BL_ROWS+RPAD]
    const int rt_stride = BL_ROWS + RPAD;
    float* y_lds = (float*)(rt_lds + BCH * rt_stride + 8);      // [BL_ROWS]
    float* red_lds = y_lds + BL_ROWS;                           // [256]

    // Theta^T staged once per block (a theta read from global inside the
    // MFMA loop poisons the vmcnt pipeline: its wait is FIFO-ordered after
    // the chunk-prefetch loads, so every chunk step drained its prefetch --
    // measured 2.97 ms vs 1.56 ms v1 before this was reverted to LDS)
    for (int idx = threadIdx.x * 8; idx < BCH * K; idx += 256 * 8) {
        const int b = idx / K;
        const int k = idx % K;
        *(U4*)&th_lds[b * th_stride + k] = *(const U4*)&theta_t[b * K + k];
    }

    typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
    typedef __attribute__((ext_vector_type(4))) float f32x4_t;
    union frag_u { bf16x8_t v; unsigned short u[8]; U4 q; };

    f32x4_t g_acc[n_chunks * 2];
#pragma unroll
    for (int t = 0; t < n_chunks * 2; ++t) g_acc[t] = (f32x4_t){0.f, 0.f, 0.f, 0.f};
    float logp_acc = 0.f;  // this lane's chain partial (chain = lane&15)

    // staging: thread owns rows {r0, r0+16, r0+32, r0+48}, one 16B group each
    const int st_r0 = threadIdx.x / 16;               // 0..15
    const int st_g = threadIdx.x % 16;                // 16B group within chunk
    const int st_key = ((st_r0 >> 3) & 1) << 1;       // swizzle key (rows +16 share it)
    const long long row_max = n_rows - 1;

    const long long n_tiles = (n_rows + BL_ROWS - 1) / BL_ROWS;
    for (long long tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
        const long long row0 = tile * BL_ROWS;
        if (threadIdx.x < BL_ROWS) {
            const long long r = row0 + threadIdx.x;
            y_lds[threadIdx.x] = r < n_rows ? bf16_bits_to_f32(y[r]) : 0.f;
        }

        U4 ld[4];
        // global loads are linear (coalesced); the XOR swizzle is applied at
        // the LDS WRITE, and compensated at both read sites.
#define LOAD_CHUNK2(c)                                                             _Pragma("unroll") for (int rr = 0; rr < 4; ++rr) {                                 const long long row = row0 + st_r0 + rr * 16;                                  ld[rr] = (U4){0, 0, 0, 0};                                                     if (row < n_rows)                                                                  ld[rr] = *(const U4*)&X[row * (long long)K + (c) * BL_CHUNK + st_g * 8];   }
#define WRITE_CHUNK2(buf)                                                          _Pragma("unroll") for (int rr = 0; rr < 4; ++rr)                                   *(U4*)&x_lds[(buf) * x_buf + (st_r0 + rr * 16) * x_stride +                                 (st_g ^ st_key) * 8] = ld[rr];

        LOAD_CHUNK2(0)
        WRITE_CHUNK2(0)
        int cur = 0;

        // ---- phase A: Z = X . Theta ----
        // v1's load placement is kept deliberately: issuing chunk c+1's
        // loads BEFORE the MFMAs forces the allocator to keep 4 separate
        // load register sets alive across them (4 loads in flight).  A
        // write-after-barrier variant (load c+2 right after the staging
        // write) let hipcc coalesce all 4 loads into ONE register quad
        // with s_waitcnt vmcnt(0) after EACH -- serial HBM round trips,
        // measured 2.81 ms vs 1.57 (see git history).
        f32x4_t z_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int c = 0; c < n_chunks; ++c) {
            __syncthreads();  // buf[cur] visible (and y_lds on c==0)
            if (c + 1 < n_chunks) LOAD_CHUNK2(c + 1)
            const int arow = wid * 16 + (lane & 15);
            const int akey = (((arow >> 3) & 1) << 1);
#pragma unroll
            for (int ks = 0; ks < BL_CHUNK / 32; ++ks) {
                frag_u a, b;
                const int ag = (ks * 4 + (lane >> 4)) ^ akey;
                a.q = *(U4*)&x_lds[cur * x_buf + arow * x_stride + ag * 8];
                const int bk = c * BL_CHUNK + ks * 32 + (lane >> 4) * 8;
                b.q = *(U4*)&th_lds[(lane & 15) * th_stride + bk];
                z_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, z_acc, 0, 0, 0);
            }
            if (c + 1 < n_chunks) WRITE_CHUNK2(cur ^ 1)
            cur ^= 1;
        }

        // ---- logp + R from Z (prefetch chunk n-1 for phase B) ----
        LOAD_CHUNK2(n_chunks - 1)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int row_in_wave = (lane >> 4) * 4 + r;
            const int row_in_tile = wid * 16 + row_in_wave;
            const long long row = row0 + row_in_tile;
            const int chain = lane & 15;
            float z = z_acc[r];
            float yv = y_lds[row_in_tile];
            float resid = 0.f;
            if (row < n_rows) {
                const float sp = fmaxf(z, 0.f) + log1pf(__expf(-fabsf(z)));
                logp_acc += yv * z - sp;
                resid = yv - 1.f / (1.f + __expf(-z));
            }
            union { float f; unsigned int u; } cv;
            cv.f = resid;
            const unsigned int rnd = 0x7fff + ((cv.u >> 16) & 1);
            rt_lds[chain * rt_stride + row_in_tile] = (unsigned short)((cv.u + rnd) >> 16);
        }
        __syncthreads();  // R complete; x_lds free
        WRITE_CHUNK2(0)   // buffer 0 <- the chunk-(n-1) data loaded above
        cur = 0;

        // ---- phase B: G += X_chunk^T . R, chunks walked in REVERSE ----
        // phase A finished on chunk n-1, so that chunk's lines are the
        // L2-hottest; walking n-1..0 maximizes phase B's L2 hit rate on the
        // re-read (forward order re-read chunk 0 first, the line most
        // likely already evicted).
#pragma unroll
        for (int ci = 0; ci < n_chunks; ++ci) {
            const int c = n_chunks - 1 - ci;
            __syncthreads();
            if (c - 1 >= 0) LOAD_CHUNK2(c - 1)
#pragma unroll
            for (int t2 = 0; t2 < 2; ++t2) {
                const int kcol0 = wid * 32 + t2 * 16;
                f32x4_t acc = g_acc[c * 2 + t2];
#pragma unroll
                for (int rs = 0; rs < 2; ++rs) {
                    frag_u a, b;
                    const int kcol = kcol0 + (lane & 15);
                    const int arow0 = rs * 32 + (lane >> 4) * 8;
                    // all 8 rows of this fragment share (row>>3), so the
                    // swizzle key (and the whole LDS column base) hoists
                    const int bkey = (((arow0 >> 3) & 1) << 1);
                    const int bg = (kcol >> 3) ^ bkey;
                    const unsigned short* col =
                        &x_lds[cur * x_buf + arow0 * x_stride + bg * 8 + (kcol & 7)];
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        a.u[j] = col[j * x_stride];
                    b.q = *(U4*)&rt_lds[(lane & 15) * rt_stride + rs * 32 + (lane >> 4) * 8];
                    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
                }
                g_acc[c * 2 + t2] = acc;
            }
            if (c - 1 >= 0) WRITE_CHUNK2(cur ^ 1)
            cur ^= 1;
        }
        __syncthreads();  // rt_lds reuse next tile
#undef LOAD_CHUNK2
#undef WRITE_CHUNK2
    }

    // ---- epilogue: block partials -> slab (same layout as v1) ----
    red_lds[threadIdx.x] = logp_acc;
    __syncthreads();
    float* slab_blk = slab + (long long)blockIdx.x * (BCH + (long long)K * BCH);
    if (threadIdx.x < BCH) {
        float s = 0.f;
        for (int i = threadIdx.x; i < 256; i += BCH) s += red_lds[i];
        slab_blk[threadIdx.x] = s;
    }
    float* g_slab = slab_blk + BCH;
#pragma unroll
    for (int c = 0; c < n_chunks; ++c) {
#pragma unroll
        for (int t2 = 0; t2 < 2; ++t2) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int kcol = c * BL_CHUNK + wid * 32 + t2 * 16 + (lane >> 4) * 4 + r;
                const int chain = lane & 15;
                g_slab[(long long)kcol * BCH + chain] = g_acc[c * 2 + t2][r];
            }
        }
    }
}

// ---------------------------------------------------------------------------
// Batched logistic v3: glds-staged, tile-RESIDENT -- phase B never re-reads
// ---------------------------------------------------------------------------
// v2 sits at the 2x-traffic floor (~1.03 ms at 2e6x1024: phase B re-reads X
// from HBM).  v3 removes the re-read: a 32-row x K tile stays resident in
// LDS across BOTH phases.  Staging is global_load_lds (one 1-KB DMA per
// 512-col row half, no VGPR round trip), pipelined one tile ahead through
// three 32-KB half-buffers with COUNTED vmcnt waits and raw s_barriers (a
// __syncthreads with a glds in flight drains the whole queue -- guide
// "Pipelining across barriers").  1 block/CU, contiguous tile ranges.
//
// Bank swizzle: the pad-free glds image (row stride 512 u16 == 0 mod 64
// dwords) would put every row on the same banks, so the SOURCE column
// group of each lane is XOR-permuted by key(row) = rotl4_by2(row & 15):
// 4-bit-injective (phase-A b128 slots stay distinct) and key(r) ^
// key(r+8) == 2 (the phase-B u16 lane-pairs that share a bank mod 8 get
// split by bit1, while bit0 stays free for the kcol-block bit).
//
// Per-wave column split is interleaved across halves (wave w owns cols
// [w*128, w*128+128) of EACH half) so both phases keep all 4 waves busy
// on whichever half is resident, and the h1 buffer can be refilled for
// tile t+1 while phase B finishes tile t's h0 columns.

#define V3_ROWS 32        // rows per tile
#define V3_HALF 512       // columns per half (K = 1024 only)

__device__ __forceinline__ int v3_key(int row) {
    const int x = row & 15;
    return ((x << 2) | (x >> 2)) & 15;  // rotl4 by 2
}

#define V3_ASM_VMCNT(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")
#define V3_BARRIER()                                                    \
    do {                                                                \
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");              \
        __builtin_amdgcn_s_barrier();                                   \
    } while (0)

// stage rows [8w, 8w+8) of one 512-col half: one 1-KB LDS-DMA per row,
// source columns pre-swizzled per lane.  dst rows are wave-uniform.
//
// The DMA is issued by INLINE ASM, not the builtin: hipcc tracks builtin
// glds in its memory model and bundles `s_waitcnt vmcnt(0)` into the
// lgkm wait of EVERY later ds_read that might alias -- which drains the
// cross-tile prefetch at each MFMA (observed in the .s).  The asm form
// leaves its completion entirely to this kernel's hand-counted
// V3_ASM_VMCNT waits (the glds queue shares vmcnt, FIFO).
__device__ __forceinline__ void v3_glds_row(const void* src, unsigned lds_byte_off) {
    // nt: each X line is streamed by exactly one CU exactly once (the
    // tile-resident design's whole point), the guide's nt-weights case
    // (issued->landed -18%).  No "memory" clobber: it would make hipcc
    // bundle a conservative vmcnt(0) into every later ds_read's wait (the
    // exact drain this asm form exists to avoid) -- ordering is carried
    // by the volatile asm barriers (which DO clobber memory).
    asm volatile(
        "s_mov_b32 m0, %0\n\t"
        "s_nop 0\n\t"
        "global_load_lds_dwordx4 %1, off nt"
        :: "s"(lds_byte_off), "v"(src));
}

__device__ __forceinline__ void v3_glds_row_cached(const void* src, unsigned lds_byte_off) {
    asm volatile(
        "s_mov_b32 m0, %0\n\t"
        "s_nop 0\n\t"
        "global_load_lds_dwordx4 %1, off"
        :: "s"(lds_byte_off), "v"(src));
}

// load one y value through a HIDDEN asm global load: it rides the same
// vmcnt queue as the tile DMAs (so the [4] counted wait covers it) and,
// unlike an ordinary tracked load, does not make hipcc insert a
// conservative vmcnt(0) drain at the use.  OOB rows clamp the address
// (uniform issue keeps the per-wave ledger exact); the caller masks the
// value.  (A 1-byte-per-lane LDS-DMA variant of this was tried first and
// produced wrong y values -- ubyte DMA lane addressing did not match the
// lane*size model; reverted to register loads.)
__device__ __forceinline__ unsigned v3_load_y_asm(const unsigned short* addr) {
    unsigned v;
    asm volatile("global_load_ushort %0, %1, off"
                 : "=v"(v) : "v"(addr));
    return v;
}

__device__ __forceinline__ void v3_stage_half(
    const unsigned short* __restrict__ X, long long n_rows, int K,
    long long row0, int colbase, unsigned short* half_buf,
    const char* smem_base, int wid, int lane, int nt_on
) {
#pragma unroll
    for (int rr = 0; rr < 8; ++rr) {
        const int r = wid * 8 + rr;
        const long long grow = row0 + r;
        unsigned short* dst = half_buf + r * V3_HALF;
        if (grow < n_rows) {
            const unsigned short* src =
                &X[grow * (long long)K + colbase + ((lane ^ v3_key(r)) * 8)];
            // the row base is wave-uniform by construction; readfirstlane
            // makes that provable so the asm "s" constraint gets an SGPR
            const unsigned off = __builtin_amdgcn_readfirstlane(
                (unsigned)((const char*)dst - smem_base));
            if (nt_on & 1) v3_glds_row(src, off);
            else v3_glds_row_cached(src, off);
        } else {
            *(U4*)&dst[lane * 8] = (U4){0, 0, 0, 0};
        }
    }
}

// v4 staging: one 1-KB DMA = 4 rows x 128 cols landing as 8 contiguous
// [4][16] subtiles; per wave 8 DMAs (row-groups 2w..2w+1 x 4 col-octets),
// so the v3 vmcnt ledgers hold unchanged.  Lane n supplies the source of
// image element n*8: subtile s = n>>3, row j = (n&7)>>1, col-half h = n&1.
__device__ __forceinline__ void v4_stage_half(
    const unsigned short* __restrict__ X, long long n_rows, int K,
    long long row0, int colbase, unsigned short* half_buf,
    const char* smem_base, int wid, int lane, int nt_on
) {
#pragma unroll
    for (int rr = 0; rr < 8; ++rr) {
        const int rg = wid * 2 + (rr >> 2);   // row-group (4 rows) 0..7
        const int oct = rr & 3;               // col-octet (128 cols) 0..3
        const long long grow0 = row0 + rg * 4;
        unsigned short* dst = half_buf + (rg * 32 + oct * 8) * 64;
        const int s = lane >> 3;
        const int j = (lane & 7) >> 1;
        const int h = lane & 1;
        if (grow0 + 4 <= n_rows) {
            const unsigned short* srcp =
                &X[(grow0 + j) * (long long)K + colbase + oct * 128 + s * 16 + h * 8];
            const unsigned off = __builtin_amdgcn_readfirstlane(
                (unsigned)((const char*)dst - smem_base));
            if (nt_on & 1) v3_glds_row(srcp, off);
            else v3_glds_row_cached(srcp, off);
        } else {
            // edge row-group: zero-fill, then guarded plain writes of the
            // in-range rows through the subtiled mapping
            *(U4*)&dst[lane * 8] = (U4){0, 0, 0, 0};
            const long long grow = grow0 + j;
            if (grow < n_rows) {
                *(U4*)&dst[(s * 64) + (j * 16) + h * 8] =
                    *(const U4*)&X[grow * (long long)K + colbase + oct * 128 +
                                   s * 16 + h * 8];
            }
        }
    }
}

// Small-K sibling of v3 (K <= 512, multiple of 128): a whole 32-row x K
// tile is ONE <=32-KB buffer, so the pipeline is a plain 3-buffer rotation
// with prefetch depth TWO tiles (the K=1024 kernel must split tiles into
// column halves to fit; here each buffer is self-sufficient, which also
// lets phase A finish Z in one sweep).  Staging, swizzle, hidden y loads
// and the counted-vmcnt discipline are identical to v3.
template <int K>
__global__ __launch_bounds__(256) void k_logistic_glm_batched_v3s(
    const unsigned short* __restrict__ X,   // [N][K] bf16
    const unsigned short* __restrict__ y,   // [N] bf16
    long long n_rows,
    const unsigned short* __restrict__ theta_t,  // [16][K] bf16
    float* __restrict__ slab,                    // [grid][16 + K*16]
    int nt_on                                    // nt on the tile DMAs (A/B)
) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    constexpr int QC = K / 4;        // columns per wave quarter
    constexpr int NKS = QC / 32;     // phase-A MFMA steps per quarter
    constexpr int NT2 = QC / 16;     // phase-B column tiles per quarter
    constexpr int TBUF = V3_ROWS * K;  // u16 per tile buffer

    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned short* th_lds = (unsigned short*)smem;        // [16][K+TPAD]
    const int th_stride = K + TPAD;
    unsigned short* x_base = th_lds + BCH * th_stride;     // 3 x [32][K]
    unsigned short* rt_lds = x_base + 3 * TBUF;            // [16][32+RPAD]
    const int rt_stride = V3_ROWS + RPAD;
    float* zc_lds = (float*)(rt_lds + BCH * rt_stride);    // [4][32][16]
    float* red_lds = zc_lds + 4 * V3_ROWS * BCH;           // [256]

    typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
    typedef __attribute__((ext_vector_type(4))) float f32x4_t;
    union frag_u { bf16x8_t v; unsigned short u[8]; U4 q; };

    for (int idx = threadIdx.x * 8; idx < BCH * K; idx += 256 * 8) {
        const int b = idx / K;
        const int k = idx % K;
        *(U4*)&th_lds[b * th_stride + k] = *(const U4*)&theta_t[b * K + k];
    }
    __syncthreads();

    f32x4_t g_acc[NT2];
#pragma unroll
    for (int t = 0; t < NT2; ++t) g_acc[t] = (f32x4_t){0.f, 0.f, 0.f, 0.f};
    float logp0 = 0.f, logp1 = 0.f;

    const long long n_tiles = (n_rows + V3_ROWS - 1) / V3_ROWS;
    const long long tpb = (n_tiles + gridDim.x - 1) / gridDim.x;
    const long long t_begin = blockIdx.x * tpb;
    const long long t_end = t_begin + tpb < n_tiles ? t_begin + tpb : n_tiles;

    // stage one whole tile into buffer `buf`: at K=512 one row is exactly
    // the 1-KB glds span (row stride = K*2 = 1024 B), so the staging shape
    // is identical to v3's; fully out-of-range rows zero-fill.
    static_assert(K == 512, "v3s is instantiated for the padded K=512 granule");
    auto stage_tile = [&](long long row0, int buf) {
#pragma unroll
        for (int rr = 0; rr < 8; ++rr) {
            const int r = wid * 8 + rr;
            const long long grow = row0 + r;
            unsigned short* dst = x_base + buf * TBUF + r * K;
            if (grow < n_rows) {
                const unsigned short* src =
                    &X[grow * (long long)K + ((lane ^ v3_key(r)) * 8)];
                const unsigned off3 = __builtin_amdgcn_readfirstlane(
                    (unsigned)((const char*)dst - smem));
                if (nt_on & 1) v3_glds_row(src, off3);
                else v3_glds_row_cached(src, off3);
            } else {
                *(U4*)&dst[lane * 8] = (U4){0, 0, 0, 0};
            }
        }
    };

    if (t_begin < t_end) {
        stage_tile(t_begin * V3_ROWS, 0);
        if (t_begin + 1 < t_end) stage_tile((t_begin + 1) * V3_ROWS, 1);

#pragma unroll 1
        for (long long tile = t_begin; tile < t_end; ++tile) {
            const long long row0 = tile * V3_ROWS;
            unsigned short* xb = x_base + (int)((tile - t_begin) % 3) * TBUF;
            const bool more1 = tile + 1 < t_end;
            const bool more2 = tile + 2 < t_end;

            // [a] own tile's DMAs complete; newer: t+1's (8/RPD) if issued
            if (more1) V3_ASM_VMCNT(8); else V3_ASM_VMCNT(0);
            V3_BARRIER();
            // y loads (hidden) BEFORE the prefetch, as in v3
            const long long ymax = n_rows - 1;
            long long yr0 = row0 + (threadIdx.x >> 4);
            long long yr1 = row0 + 16 + (threadIdx.x >> 4);
            const unsigned yb0 = v3_load_y_asm(y + (yr0 > ymax ? ymax : yr0));
            const unsigned yb1 = v3_load_y_asm(y + (yr1 > ymax ? ymax : yr1));
            if (more2) stage_tile((tile + 2) * V3_ROWS, (int)((tile + 2 - t_begin) % 3));

            // ---- phase A: complete Z over this wave's K-quarter ----
            f32x4_t z0 = {0.f, 0.f, 0.f, 0.f}, z1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int ks = 0; ks < NKS; ++ks) {
                const int kc = wid * QC + ks * 32;
                frag_u bf, a0, a1;
                const int g = (kc >> 3) + (lane >> 4);
                const int arow0 = lane & 15;
                const int arow1 = 16 + (lane & 15);
                a0.q = *(U4*)&xb[arow0 * K + ((g ^ v3_key(arow0)) * 8)];
                a1.q = *(U4*)&xb[arow1 * K + ((g ^ v3_key(arow1)) * 8)];
                const int bk = kc + (lane >> 4) * 8;
                bf.q = *(U4*)&th_lds[(lane & 15) * th_stride + bk];
                z0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0.v, bf.v, z0, 0, 0, 0);
                z1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1.v, bf.v, z1, 0, 0, 0);
            }
            {
                float* zc = zc_lds + wid * V3_ROWS * BCH;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    zc[((lane >> 4) * 4 + r) * BCH + (lane & 15)] = z0[r];
                    zc[(16 + (lane >> 4) * 4 + r) * BCH + (lane & 15)] = z1[r];
                }
            }
            // y complete: newer ops are the prefetch's DMAs only
            if (more2) V3_ASM_VMCNT(8); else V3_ASM_VMCNT(0);
            V3_BARRIER();
            {
#pragma unroll
                for (int s = 0; s < 2; ++s) {
                    const int slot = threadIdx.x + s * 256;
                    const int row = slot >> 4;
                    const int chain = slot & 15;
                    float z = 0.f;
#pragma unroll
                    for (int w = 0; w < 4; ++w)
                        z += zc_lds[(w * V3_ROWS + row) * BCH + chain];
                    const long long grow = row0 + row;
                    float resid = 0.f;
                    if (grow < n_rows) {
                        const float yv = bf16_bits_to_f32(
                            (unsigned short)(s == 0 ? yb0 : yb1));
                        const float sp = fmaxf(z, 0.f) + log1pf(__expf(-fabsf(z)));
                        const float term = yv * z - sp;
                        if (s == 0) logp0 += term; else logp1 += term;
                        resid = yv - 1.f / (1.f + __expf(-z));
                    }
                    union { float f; unsigned int u; } cv;
                    cv.f = resid;
                    const unsigned int rnd = 0x7fff + ((cv.u >> 16) & 1);
                    rt_lds[chain * rt_stride + row] =
                        (unsigned short)((cv.u + rnd) >> 16);
                }
            }
            V3_BARRIER();  // R visible

            // ---- phase B over this wave's quarter ----
#pragma unroll
            for (int t2 = 0; t2 < NT2; ++t2) {
                const int kc = wid * QC + t2 * 16;
                const int kcol = kc + (lane & 15);
                f32x4_t acc = g_acc[t2];
                frag_u a, b;
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    const int row = (lane >> 4) * 8 + j;
                    a.u[j] = xb[row * K +
                                (((kcol >> 3) ^ v3_key(row)) * 8) + (kcol & 7)];
                }
                b.q = *(U4*)&rt_lds[(lane & 15) * rt_stride + (lane >> 4) * 8];
                g_acc[t2] =
                    __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
            }
            V3_BARRIER();  // xb free for the t+3 prefetch next iteration
        }
    }

    // ---- epilogue ----
    __syncthreads();
    red_lds[threadIdx.x] = logp0 + logp1;
    __syncthreads();
    float* slab_blk = slab + (long long)blockIdx.x * (BCH + (long long)K * BCH);
    if (threadIdx.x < BCH) {
        float s = 0.f;
        for (int i = threadIdx.x; i < 256; i += BCH) s += red_lds[i];
        slab_blk[threadIdx.x] = s;
    }
    float* g_slab = slab_blk + BCH;
#pragma unroll
    for (int t2 = 0; t2 < NT2; ++t2) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int kcol = wid * QC + t2 * 16 + (lane >> 4) * 4 + r;
            const int chain = lane & 15;
            g_slab[(long long)kcol * BCH + chain] = g_acc[t2][r];
        }
    }
}

// v4 (TRB=1) image: X stored as row-major [4][16] subtiles so phase-B
// fragments come from gfx950's hardware transpose-read (probe-verified
// semantics, scripts/tr_probe.hip + gpurun_out/tr_probe.txt: each 16-lane
// group hands lane (l&15) COLUMN l&15 of the 64-elem tile its addresses
// cover, elems {base + (l&15) + 16j}).  Element offset of (row, col)
// within a 32-row x 512-col half:
__device__ __forceinline__ int v4_img(int row, int col) {
    return (((row >> 2) * 32 + (col >> 4)) << 6) + ((row & 3) << 4) + (col & 15);
}

template <int PROF, int TRB>
__global__ __launch_bounds__(256) void k_logistic_glm_batched_v3(
    const unsigned short* __restrict__ X,   // [N][1024] bf16
    const unsigned short* __restrict__ y,   // [N] bf16
    long long n_rows,
    const unsigned short* __restrict__ theta_t,  // [16][1024] bf16
    float* __restrict__ slab,                    // [grid][16 + 1024*16]
    int nt_on                                    // bit0: nt DMAs; bit1: profile
) {
    constexpr int K = 2 * V3_HALF;
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    // FED_V3_PROF: wid-0 lane-0 of every block accumulates cycle counts
    // for the two stage-gate waits, the mid (zc/R) barrier region and the
    // phase-B h1 barrier into slab rows past the v3 grid (256 blocks x 8
    // u64 in otherwise-unused workspace) -- splits PMC's "35% wait" into
    // stage-starvation vs barrier-skew without touching the default path.
    constexpr int prof_on = PROF;  // compile-time: the default instantiation
                                   // carries ZERO instrumentation branches
    unsigned long long p_gate1 = 0, p_gate4 = 0, p_mid = 0, p_bh = 0, p_t0 = 0;
    unsigned long long p_phA = 0, p_phB = 0;
    if (prof_on && wid == 0 && lane == 0) p_t0 = __builtin_amdgcn_s_memtime();

    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned short* th_lds = (unsigned short*)smem;        // [16][K+TPAD]
    const int th_stride = K + TPAD;
    // three half buffers live at x_base + {0,1,2}*HBUF; they are addressed
    // by OFFSET so every access provably derives from the __shared__ base
    // (a pointer array indexed at runtime loses the address-space
    // inference and hipcc emits flat_load for what should be ds_read)
    constexpr int HBUF = V3_ROWS * V3_HALF;
    unsigned short* x_base = th_lds + BCH * th_stride;     // 3 x [32][512]
    unsigned short* rt_lds = x_base + 3 * HBUF;            // [16][32+RPAD]
    const int rt_stride = V3_ROWS + RPAD;
    float* zc_lds = (float*)(rt_lds + BCH * rt_stride);    // [4][32][16]
    float* red_lds = zc_lds + 4 * V3_ROWS * BCH;           // [256]

    typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
    typedef __attribute__((ext_vector_type(4))) float f32x4_t;
    union frag_u { bf16x8_t v; unsigned short u[8]; U4 q; };

    // ---- stage Theta^T once (ordinary loads; BEFORE any glds) ----
    for (int idx = threadIdx.x * 8; idx < BCH * K; idx += 256 * 8) {
        const int b = idx / K;
        const int k = idx % K;
        *(U4*)&th_lds[b * th_stride + k] = *(const U4*)&theta_t[b * K + k];
    }
    __syncthreads();

    f32x4_t g_acc[16];
#pragma unroll
    for (int t = 0; t < 16; ++t) g_acc[t] = (f32x4_t){0.f, 0.f, 0.f, 0.f};
    float logp0 = 0.f, logp1 = 0.f;  // this thread's two (row, chain) slots

    // contiguous tile range per block (L2 locality + equal glds ledgers)
    const long long n_tiles = (n_rows + V3_ROWS - 1) / V3_ROWS;
    const long long tpb = (n_tiles + gridDim.x - 1) / gridDim.x;
    const long long t_begin = blockIdx.x * tpb;
    const long long t_end = t_begin + tpb < n_tiles ? t_begin + tpb : n_tiles;

    if (t_begin < t_end) {
        // prologue: stage tile t_begin fully.  h0 buffers alternate 0/2,
        // h1 lives in buffer 1.
        if constexpr (TRB) {
            v4_stage_half(X, n_rows, K, t_begin * V3_ROWS, 0, x_base, smem, wid, lane, nt_on);
            v4_stage_half(X, n_rows, K, t_begin * V3_ROWS, V3_HALF, x_base + HBUF, smem, wid, lane, nt_on);
        } else {
            v3_stage_half(X, n_rows, K, t_begin * V3_ROWS, 0, x_base, smem, wid, lane, nt_on);
            v3_stage_half(X, n_rows, K, t_begin * V3_ROWS, V3_HALF, x_base + HBUF, smem, wid, lane, nt_on);
        }

        int h0sel = 0;  // buffer index (0 or 2) holding the CURRENT tile's h0
#pragma unroll 1
        for (long long tile = t_begin; tile < t_end; ++tile) {
            const long long row0 = tile * V3_ROWS;
            unsigned short* h0 = x_base + h0sel * HBUF;
            unsigned short* h1 = x_base + HBUF;
            unsigned short* h0n = x_base + (h0sel ^ 2) * HBUF;  // next tile's h0
            const bool more = tile + 1 < t_end;

            // [1] own h0 DMAs done (h1's 8 may stay in flight), all waves
            unsigned long long pt = 0;
            if (prof_on && wid == 0 && lane == 0) pt = __builtin_amdgcn_s_memtime();
            V3_ASM_VMCNT(8);
            V3_BARRIER();
            if (prof_on && wid == 0 && lane == 0)
                p_gate1 += __builtin_amdgcn_s_memtime() - pt;
            // this thread's two y values (R-step slots), hidden loads on
            // the same queue, issued BEFORE the h0 prefetch so the [4]
            // counted wait (which leaves only the prefetch outstanding)
            // provably covers them
            const long long ymax = n_rows - 1;
            long long yr0 = row0 + (threadIdx.x >> 4);
            long long yr1 = row0 + 16 + (threadIdx.x >> 4);
            const unsigned yb0 = v3_load_y_asm(y + (yr0 > ymax ? ymax : yr0));
            const unsigned yb1 = v3_load_y_asm(y + (yr1 > ymax ? ymax : yr1));
            // [2] prefetch next tile's h0 as deep as possible
            if (more) {
                if constexpr (TRB)
                    v4_stage_half(X, n_rows, K, row0 + V3_ROWS, 0, h0n, smem, wid, lane, nt_on);
                else
                    v3_stage_half(X, n_rows, K, row0 + V3_ROWS, 0, h0n, smem, wid, lane, nt_on);
            }

            // ---- phase A on h0: z += X[:, w*128 .. +128) . theta ----
            unsigned long long pa = 0;
            if (prof_on && wid == 0 && lane == 0) pa = __builtin_amdgcn_s_memtime();
            f32x4_t z0 = {0.f, 0.f, 0.f, 0.f}, z1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int ks = 0; ks < 4; ++ks) {
                const int kc = wid * 128 + ks * 32;  // column in half
                frag_u b0f, b1f, a0, a1;
                const int g = (kc >> 3) + (lane >> 4);  // 16B group in half
                const int arow0 = lane & 15;            // rows 0..15
                const int arow1 = 16 + (lane & 15);     // rows 16..31
                if constexpr (TRB) {
                    a0.q = *(U4*)&h0[v4_img(arow0, g * 8)];
                    a1.q = *(U4*)&h0[v4_img(arow1, g * 8)];
                } else {
                    a0.q = *(U4*)&h0[arow0 * V3_HALF + ((g ^ v3_key(arow0)) * 8)];
                    a1.q = *(U4*)&h0[arow1 * V3_HALF + ((g ^ v3_key(arow1)) * 8)];
                }
                const int bk = kc + (lane >> 4) * 8;    // theta col (h0)
                b0f.q = *(U4*)&th_lds[(lane & 15) * th_stride + bk];
                z0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0.v, b0f.v, z0, 0, 0, 0);
                b1f.q = b0f.q;
                z1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1.v, b1f.v, z1, 0, 0, 0);
            }
            if (prof_on && wid == 0 && lane == 0)
                p_phA += __builtin_amdgcn_s_memtime() - pa;
            // [4] own h1 DMAs + y loads done.  The count must match what
            // was actually issued after them: 8 prefetch DMAs on interior
            // tiles, NOTHING on the last tile (an unconditional vmcnt(8)
            // there would leave h1/y unwaited -- a timing-dependent race).
            if (prof_on && wid == 0 && lane == 0) pt = __builtin_amdgcn_s_memtime();
            if (more) V3_ASM_VMCNT(8); else V3_ASM_VMCNT(0);
            V3_BARRIER();
            if (prof_on && wid == 0 && lane == 0)
                p_gate4 += __builtin_amdgcn_s_memtime() - pt;
            if (prof_on && wid == 0 && lane == 0) pa = __builtin_amdgcn_s_memtime();
            // ---- phase A on h1 ----
#pragma unroll
            for (int ks = 0; ks < 4; ++ks) {
                const int kc = wid * 128 + ks * 32;
                frag_u bf, a0, a1;
                const int g = (kc >> 3) + (lane >> 4);
                const int arow0 = lane & 15;
                const int arow1 = 16 + (lane & 15);
                if constexpr (TRB) {
                    a0.q = *(U4*)&h1[v4_img(arow0, g * 8)];
                    a1.q = *(U4*)&h1[v4_img(arow1, g * 8)];
                } else {
                    a0.q = *(U4*)&h1[arow0 * V3_HALF + ((g ^ v3_key(arow0)) * 8)];
                    a1.q = *(U4*)&h1[arow1 * V3_HALF + ((g ^ v3_key(arow1)) * 8)];
                }
                const int bk = V3_HALF + kc + (lane >> 4) * 8;  // theta col (h1)
                bf.q = *(U4*)&th_lds[(lane & 15) * th_stride + bk];
                z0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0.v, bf.v, z0, 0, 0, 0);
                z1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1.v, bf.v, z1, 0, 0, 0);
            }
            if (prof_on && wid == 0 && lane == 0)
                p_phA += __builtin_amdgcn_s_memtime() - pa;
            // ---- combine the 4 waves' z quarters; logp + R ----
            {
                float* zc = zc_lds + wid * V3_ROWS * BCH;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    zc[((lane >> 4) * 4 + r) * BCH + (lane & 15)] = z0[r];
                    zc[(16 + (lane >> 4) * 4 + r) * BCH + (lane & 15)] = z1[r];
                }
            }
            if (prof_on && wid == 0 && lane == 0) pt = __builtin_amdgcn_s_memtime();
            V3_BARRIER();
            if (prof_on && wid == 0 && lane == 0)
                p_mid += __builtin_amdgcn_s_memtime() - pt;
            {
#pragma unroll
                for (int s = 0; s < 2; ++s) {
                    const int slot = threadIdx.x + s * 256;
                    const int row = slot >> 4;      // 0..31
                    const int chain = slot & 15;
                    float z = 0.f;
#pragma unroll
                    for (int w = 0; w < 4; ++w)
                        z += zc_lds[(w * V3_ROWS + row) * BCH + chain];
                    const long long grow = row0 + row;
                    float resid = 0.f;
                    if (grow < n_rows) {
                        const float yv = bf16_bits_to_f32(
                            (unsigned short)(s == 0 ? yb0 : yb1));
                        const float sp = fmaxf(z, 0.f) + log1pf(__expf(-fabsf(z)));
                        const float term = yv * z - sp;
                        if (s == 0) logp0 += term; else logp1 += term;
                        resid = yv - 1.f / (1.f + __expf(-z));
                    }
                    union { float f; unsigned int u; } cv;
                    cv.f = resid;
                    const unsigned int rnd = 0x7fff + ((cv.u >> 16) & 1);
                    rt_lds[chain * rt_stride + row] =
                        (unsigned short)((cv.u + rnd) >> 16);
                }
            }
            if (prof_on && wid == 0 && lane == 0) pt = __builtin_amdgcn_s_memtime();
            V3_BARRIER();  // R visible
            if (prof_on && wid == 0 && lane == 0)
                p_mid += __builtin_amdgcn_s_memtime() - pt;

            if (prof_on && wid == 0 && lane == 0) pa = __builtin_amdgcn_s_memtime();
            // ---- phase B, h1 columns first (frees h1 for the refill) ----
#pragma unroll
            for (int t2 = 0; t2 < 8; ++t2) {
                const int kc = wid * 128 + t2 * 16;       // column in half
                const int kcol = kc + (lane & 15);
                f32x4_t acc = g_acc[8 + t2];
                frag_u a, b;
                if constexpr (TRB) {
                    // two hardware transpose reads: lane (l&15) gets kcol
                    // column kc+(l&15) for rows (l>>4)*8..+3 and +4..+7
                    const unsigned base0 = (unsigned)((const char*)&h1[
                        v4_img((lane >> 4) * 8, kc)] - smem) + (lane & 15) * 8u;
                    const unsigned base1 = (unsigned)((const char*)&h1[
                        v4_img((lane >> 4) * 8 + 4, kc)] - smem) + (lane & 15) * 8u;
                    unsigned long long v0, v1;
                    asm volatile(
                        "ds_read_b64_tr_b16 %0, %2\n\t"
                        "ds_read_b64_tr_b16 %1, %3\n\t"
                        "s_waitcnt lgkmcnt(0)"
                        : "=v"(v0), "=v"(v1) : "v"(base0), "v"(base1) : "memory");
                    ((unsigned long long*)a.u)[0] = v0;
                    ((unsigned long long*)a.u)[1] = v1;
                } else {
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        const int row = (lane >> 4) * 8 + j;
                        a.u[j] = h1[row * V3_HALF +
                                    (((kcol >> 3) ^ v3_key(row)) * 8) + (kcol & 7)];
                    }
                }
                b.q = *(U4*)&rt_lds[(lane & 15) * rt_stride + (lane >> 4) * 8];
                g_acc[8 + t2] =
                    __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
            }
            if (prof_on && wid == 0 && lane == 0) {
                p_phB += __builtin_amdgcn_s_memtime() - pa;
                pt = __builtin_amdgcn_s_memtime();
            }
            V3_BARRIER();  // everyone done reading h1
            if (prof_on && wid == 0 && lane == 0)
                p_bh += __builtin_amdgcn_s_memtime() - pt;
            // [6] refill h1 with the NEXT tile's second half
            if (more) {
                if constexpr (TRB)
                    v4_stage_half(X, n_rows, K, row0 + V3_ROWS, V3_HALF, h1, smem, wid, lane, nt_on);
                else
                    v3_stage_half(X, n_rows, K, row0 + V3_ROWS, V3_HALF, h1, smem, wid, lane, nt_on);
            }

            if (prof_on && wid == 0 && lane == 0) pa = __builtin_amdgcn_s_memtime();
            // ---- phase B, h0 columns ----
#pragma unroll
            for (int t2 = 0; t2 < 8; ++t2) {
                const int kc = wid * 128 + t2 * 16;
                const int kcol = kc + (lane & 15);
                f32x4_t acc = g_acc[t2];
                frag_u a, b;
                if constexpr (TRB) {
                    // two hardware transpose reads: lane (l&15) gets kcol
                    // column kc+(l&15) for rows (l>>4)*8..+3 and +4..+7
                    const unsigned base0 = (unsigned)((const char*)&h0[
                        v4_img((lane >> 4) * 8, kc)] - smem) + (lane & 15) * 8u;
                    const unsigned base1 = (unsigned)((const char*)&h0[
                        v4_img((lane >> 4) * 8 + 4, kc)] - smem) + (lane & 15) * 8u;
                    unsigned long long v0, v1;
                    asm volatile(
                        "ds_read_b64_tr_b16 %0, %2\n\t"
                        "ds_read_b64_tr_b16 %1, %3\n\t"
                        "s_waitcnt lgkmcnt(0)"
                        : "=v"(v0), "=v"(v1) : "v"(base0), "v"(base1) : "memory");
                    ((unsigned long long*)a.u)[0] = v0;
                    ((unsigned long long*)a.u)[1] = v1;
                } else {
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        const int row = (lane >> 4) * 8 + j;
                        a.u[j] = h0[row * V3_HALF +
                                    (((kcol >> 3) ^ v3_key(row)) * 8) + (kcol & 7)];
                    }
                }
                b.q = *(U4*)&rt_lds[(lane & 15) * rt_stride + (lane >> 4) * 8];
                g_acc[t2] =
                    __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
            }
            if (prof_on && wid == 0 && lane == 0)
                p_phB += __builtin_amdgcn_s_memtime() - pa;
            h0sel ^= 2;
        }
    }

    if (prof_on && wid == 0 && lane == 0) {
        // rows >= 300 of the workspace slab are unused at v3's grid (256)
        unsigned long long* prof =
            (unsigned long long*)(slab + 300LL * (BCH + (long long)K * BCH)) +
            8LL * blockIdx.x;
        prof[0] = p_gate1;
        prof[1] = p_gate4;
        prof[2] = p_mid;
        prof[3] = p_bh;
        prof[4] = __builtin_amdgcn_s_memtime() - p_t0;
        prof[5] = p_phA;
        prof[6] = p_phB;
    }
    // ---- epilogue: block partials -> slab (layout shared with v1/v2) ----
    __syncthreads();
    red_lds[threadIdx.x] = logp0 + logp1;
    // the two slots of one thread are chains (t&15) and ((t+256)&15) == same
    // chain, different rows -- so the plain per-chain strided sum works
    __syncthreads();
    float* slab_blk = slab + (long long)blockIdx.x * (BCH + (long long)K * BCH);
    if (threadIdx.x < BCH) {
        float s = 0.f;
        for (int i = threadIdx.x; i < 256; i += BCH) s += red_lds[i];
        slab_blk[threadIdx.x] = s;
    }
    float* g_slab = slab_blk + BCH;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
#pragma unroll
        for (int t2 = 0; t2 < 8; ++t2) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int kcol = half * V3_HALF + wid * 128 + t2 * 16 +
                                 (lane >> 4) * 4 + r;
                const int chain = lane & 15;
                g_slab[(long long)kcol * BCH + chain] = g_acc[half * 8 + t2][r];
            }
        }
    }
}

// ---------------------------------------------------------------------------
// Batched logistic, LDS-resident-tile variant: phase B never re-reads HBM
// ---------------------------------------------------------------------------
// The chunked variant's phase-B re-read misses L2 (2 blocks/CU x 32 CUs x
// 128 KB tiles = 8 MB active per 4 MB XCD L2) -> ~2x HBM traffic.  Here ONE
// block per CU holds its whole [64][K] tile in LDS (132 KB at K=1024),
// stages it once with a deep load pipeline, and runs both MFMA phases out
// of LDS.  Theta fragments read straight from L2 (32 KB, resident).

template <int K>
__global__ __launch_bounds__(256, 1) void k_logistic_glm_batched_lds(
    const unsigned short* __restrict__ X,   // [N][K] bf16
    const unsigned short* __restrict__ y,   // [N] bf16
    long long n_rows,
    const unsigned short* __restrict__ theta_t,  // [BCH][K] bf16
    float* __restrict__ slab                     // [grid][BCH + K*BCH]
) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    constexpr int n_chunks = K / BL_CHUNK;
    constexpr int x_stride = K + XPAD;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned short* x_lds = (unsigned short*)smem;              // [BL_ROWS][K+XPAD]
    unsigned short* rt_lds = x_lds + BL_ROWS * x_stride;        // [BCH][BL_ROWS+RPAD]
    const int rt_stride = BL_ROWS + RPAD;
    float* y_lds = (float*)(rt_lds + BCH * rt_stride + 8);      // [BL_ROWS]
    float* red_lds = y_lds + BL_ROWS;                           // [256]

    typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
    typedef __attribute__((ext_vector_type(4))) float f32x4_t;
    union frag_u { bf16x8_t v; unsigned short u[8]; U4 q; };

    f32x4_t g_acc[n_chunks * 2];
#pragma unroll
    for (int t = 0; t < n_chunks * 2; ++t) g_acc[t] = (f32x4_t){0.f, 0.f, 0.f, 0.f};
    float logp_acc = 0.f;

    const long long n_tiles = (n_rows + BL_ROWS - 1) / BL_ROWS;
    for (long long tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
        const long long row0 = tile * BL_ROWS;
        if (threadIdx.x < BL_ROWS) {
            const long long r = row0 + threadIdx.x;
            y_lds[threadIdx.x] = r < n_rows ? bf16_bits_to_f32(y[r]) : 0.f;
        }
        // ---- stage the WHOLE tile: 512 B per thread, 32 deep loads ----
        {
            const int r = threadIdx.x / 16;          // owns rows r, r+16, ...
            const int k0 = (threadIdx.x % 16) * 8;   // k column start
#pragma unroll
            for (int rr = 0; rr < 4; ++rr) {
                const long long row = row0 + r + rr * 16;
#pragma unroll
                for (int kk = 0; kk < K / 128; ++kk) {
                    U4 val = {0, 0, 0, 0};
                    if (row < n_rows)
                        val = *(const U4*)&X[row * (long long)K + kk * 128 + k0];
                    *(U4*)&x_lds[(r + rr * 16) * x_stride + kk * 128 + k0] = val;
                }
            }
        }
        __syncthreads();

        // ---- phase A: Z (LDS x, L2 theta) ----
        f32x4_t z_acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ks = 0; ks < K / 32; ++ks) {
            frag_u a, b;
            const int arow = wid * 16 + (lane & 15);
            const int ak = ks * 32 + (lane >> 4) * 8;
            a.q = *(U4*)&x_lds[arow * x_stride + ak];
            b.q = *(const U4*)&theta_t[(lane & 15) * K + ak];
            z_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, z_acc, 0, 0, 0);
        }

        // ---- logp + R ----
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int row_in_wave = (lane >> 4) * 4 + r;
            const int row_in_tile = wid * 16 + row_in_wave;
            const long long row = row0 + row_in_tile;
            const int chain = lane & 15;
            float z = z_acc[r];
            float yv = y_lds[row_in_tile];
            float resid = 0.f;
            if (row < n_rows) {
                const float sp = fmaxf(z, 0.f) + log1pf(__expf(-fabsf(z)));
                logp_acc += yv * z - sp;
                resid = yv - 1.f / (1.f + __expf(-z));
            }
            union { float f; unsigned int u; } cv;
            cv.f = resid;
            const unsigned int rnd = 0x7fff + ((cv.u >> 16) & 1);
            rt_lds[chain * rt_stride + row_in_tile] = (unsigned short)((cv.u + rnd) >> 16);
        }
        __syncthreads();  // R complete

        // ---- phase B: G += X^T R (all LDS) ----
#pragma unroll
        for (int c = 0; c < n_chunks; ++c) {
#pragma unroll
            for (int t2 = 0; t2 < 2; ++t2) {
                const int kcol0 = c * BL_CHUNK + wid * 32 + t2 * 16;
                f32x4_t acc = g_acc[c * 2 + t2];
#pragma unroll
                for (int rs = 0; rs < 2; ++rs) {
                    frag_u a, b;
                    const int kcol = kcol0 + (lane & 15);
                    const int arow0 = rs * 32 + (lane >> 4) * 8;
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        a.u[j] = x_lds[(arow0 + j) * x_stride + kcol];
                    b.q = *(U4*)&rt_lds[(lane & 15) * rt_stride + rs * 32 + (lane >> 4) * 8];
                    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
                }
                g_acc[c * 2 + t2] = acc;
            }
        }
        __syncthreads();  // x_lds/rt_lds reuse next tile
    }

    // ---- epilogue (same slab layout as the chunked variant) ----
    red_lds[threadIdx.x] = logp_acc;
    __syncthreads();
    float* slab_blk = slab + (long long)blockIdx.x * (BCH + (long long)K * BCH);
    if (threadIdx.x < BCH) {
        float s = 0.f;
        for (int i = threadIdx.x; i < 256; i += BCH) s += red_lds[i];
        slab_blk[threadIdx.x] = s;
    }
    float* g_slab = slab_blk + BCH;
#pragma unroll
    for (int c = 0; c < n_chunks; ++c) {
#pragma unroll
        for (int t2 = 0; t2 < 2; ++t2) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int kcol = c * BL_CHUNK + wid * 32 + t2 * 16 + (lane >> 4) * 4 + r;
                const int chain = lane & 15;
                g_slab[(long long)kcol * BCH + chain] = g_acc[c * 2 + t2][r];
            }
        }
    }
}

extern "C" int fed_logistic_glm_batched(
    const void* X, const void* y, long long n_rows, int K,
    const void* theta_t_bf16,  // [16][K] bf16, transposed theta
    double* out,               // fp64[16 + K*16]: [logp[16] | G[K][16]]
    float* workspace, long long ws_bytes,
    void* stream_v
) {
    hipStream_t stream = (hipStream_t)stream_v;
    if (K != 512 && K != 1024) return -4;
    const int block = 256;
    const char* v1_env = getenv("FED_BATCHED_V1");
    const bool v1_variant = v1_env && atoi(v1_env) != 0;
    // v2 runs 2 blocks/CU (71 KB LDS); 512 blocks covers every CU twice.
    // FED_BATCHED_GRID overrides for the L2-residency A/B: at 256 (1
    // block/CU) a block's whole 128 KB tile fits its XCD L2 share, so
    // phase B's re-read can hit L2 instead of HBM -- at the cost of half
    // the waves hiding phase A's stream latency.
    int grid_cap = v1_variant ? 304 : 512;
    if (const char* ge = getenv("FED_BATCHED_GRID")) {
        int v = atoi(ge);
        if (v >= 16 && v <= 768) grid_cap = v;
    }
    int grid = pick_grid(n_rows / BL_ROWS + 1, 1);
    if (grid > grid_cap) grid = grid_cap;
    const long long slab_cols = BCH + (long long)K * BCH;
    if ((long long)grid * slab_cols * 4 > ws_bytes)
        grid = (int)(ws_bytes / (slab_cols * 4));
    if (grid < 1) return -3;
    const int lds_bytes =
        (BCH * (K + TPAD) + 2 * BL_ROWS * (BL_CHUNK + XPAD) + BCH * (BL_ROWS + RPAD) + 8) * 2 +
        (BL_ROWS + 256) * 4 + 64;
    const int lds_bytes_v2 =
        (BCH * (K + TPAD) + 2 * BL_ROWS * (BL_CHUNK + XPAD) + BCH * (BL_ROWS + RPAD) + 8) * 2 +
        (BL_ROWS + 256) * 4 + 64;
    const char* lds_env = getenv("FED_BATCHED_LDS");
    const bool lds_variant = lds_env && atoi(lds_env) != 0;
    // v3 (glds tile-resident, 1x HBM traffic) is the default at K=1024:
    // 0.878 ms vs v2's 1.035 / v1's 1.568 at 2e6x1024x16, 5.17 ms at the
    // config-4 shard (profiles/raw_r2/r2c10_*).  At K=512 the chunked v2
    // stays default: its 64-KB tiles already re-read through L2 (measured
    // 0.567 ms ~= the 1x floor vs v3s 0.645 at 2e6x512 -- raw_r2/r2c12);
    // FED_BATCHED_V3=1/0 forces either way.
    const char* v3_env = getenv("FED_BATCHED_V3");
    const bool v3_on = v3_env ? atoi(v3_env) != 0 : (K == 1024);
    if (v3_on && !v1_variant && !lds_variant && (K == 1024 || K == 512)) {
        // glds tile-resident variant: 1 block/CU, contiguous tile ranges
        int g3 = grid;
        if (g3 > 256) g3 = 256;
        const int xbytes = K == 1024 ? 3 * V3_ROWS * V3_HALF : 3 * V3_ROWS * 512;
        const int lds3 = (BCH * (K + TPAD) + xbytes +
                          BCH * (V3_ROWS + RPAD)) * 2 +
                         (4 * V3_ROWS * BCH + 256) * 4 + 64;
        const char* nt = getenv("FED_V3_NT");
        int nt_on = nt ? (atoi(nt) != 0) : 1;  // stream-once data: nt default
        const char* pf = getenv("FED_V3_PROF");
        if (pf && atoi(pf) != 0) nt_on |= 2;
        if (K == 1024) {
            // tr_b16 phase-B image is the default: same-box A/B measured
            // 0.888-0.891 vs 0.952-0.963 ms (3 reps, profiles raw_r2/v4_*);
            // FED_BATCHED_V4=0 restores the scalar-gather image
            const char* v4e = getenv("FED_BATCHED_V4");
            if (!v4e || atoi(v4e) != 0)
                hipLaunchKernelGGL((k_logistic_glm_batched_v3<0, 1>), dim3(g3),
                                   dim3(block), lds3, stream, (const unsigned short*)X,
                                   (const unsigned short*)y, n_rows,
                                   (const unsigned short*)theta_t_bf16, workspace, nt_on);
            else if (nt_on & 2)
                hipLaunchKernelGGL((k_logistic_glm_batched_v3<1, 0>), dim3(g3),
                                   dim3(block), lds3, stream, (const unsigned short*)X,
                                   (const unsigned short*)y, n_rows,
                                   (const unsigned short*)theta_t_bf16, workspace, nt_on);
            else
                hipLaunchKernelGGL((k_logistic_glm_batched_v3<0, 0>), dim3(g3),
                                   dim3(block), lds3, stream, (const unsigned short*)X,
                                   (const unsigned short*)y, n_rows,
                                   (const unsigned short*)theta_t_bf16, workspace, nt_on);
        }
        else
            hipLaunchKernelGGL(k_logistic_glm_batched_v3s<512>, dim3(g3), dim3(block),
                               lds3, stream, (const unsigned short*)X,
                               (const unsigned short*)y, n_rows,
                               (const unsigned short*)theta_t_bf16, workspace, nt_on);
        hipError_t verr = hipGetLastError();
        if (verr != hipSuccess) return (int)verr;
        const int rg = ((int)slab_cols + 255) / 256;
        int ch3 = g3 / 8;
        if (ch3 < 1) ch3 = 1;
        hipError_t m3 = hipMemsetAsync(out, 0, slab_cols * 8, stream);
        if (m3 != hipSuccess) return (int)m3;
        hipLaunchKernelGGL(k_colsum_reduce, dim3(rg, ch3), dim3(256), 0, stream,
                           workspace, g3, (int)slab_cols, out);
        return (int)hipGetLastError();
    }
    if (!lds_variant && !v1_variant) {
        if (K == 1024)
            hipLaunchKernelGGL(k_logistic_glm_batched_v2<1024>, dim3(grid), dim3(block),
                               lds_bytes_v2, stream, (const unsigned short*)X,
                               (const unsigned short*)y, n_rows,
                               (const unsigned short*)theta_t_bf16, workspace);
        else
            hipLaunchKernelGGL(k_logistic_glm_batched_v2<512>, dim3(grid), dim3(block),
                               lds_bytes_v2, stream, (const unsigned short*)X,
                               (const unsigned short*)y, n_rows,
                               (const unsigned short*)theta_t_bf16, workspace);
    } else if (lds_variant) {
        // whole-tile-resident variant: 1 block/CU, dynamic LDS ~132 KB
        const int lds2 = (BL_ROWS * (K + XPAD) + BCH * (BL_ROWS + RPAD) + 8) * 2 +
                         (BL_ROWS + 256) * 4 + 64;
        if (grid > 304) grid = 304;
        if (K == 1024)
            hipLaunchKernelGGL(k_logistic_glm_batched_lds<1024>, dim3(grid), dim3(block), lds2,
                               stream, (const unsigned short*)X, (const unsigned short*)y,
                               n_rows, (const unsigned short*)theta_t_bf16, workspace);
        else
            hipLaunchKernelGGL(k_logistic_glm_batched_lds<512>, dim3(grid), dim3(block), lds2,
                               stream, (const unsigned short*)X, (const unsigned short*)y,
                               n_rows, (const unsigned short*)theta_t_bf16, workspace);
    } else if (K == 1024)
        hipLaunchKernelGGL(k_logistic_glm_batched<1024>, dim3(grid), dim3(block), lds_bytes,
                           stream, (const unsigned short*)X, (const unsigned short*)y, n_rows,
                           (const unsigned short*)theta_t_bf16, workspace);
    else
        hipLaunchKernelGGL(k_logistic_glm_batched<512>, dim3(grid), dim3(block), lds_bytes,
                           stream, (const unsigned short*)X, (const unsigned short*)y, n_rows,
                           (const unsigned short*)theta_t_bf16, workspace);
    hipError_t kerr = hipGetLastError();
    if (kerr != hipSuccess) return (int)kerr;
    const int rgrid = ((int)slab_cols + 255) / 256;
    int chunks = grid / 8;
    if (chunks < 1) chunks = 1;
    if (chunks > 64) chunks = 64;
    hipError_t merr = hipMemsetAsync(out, 0, slab_cols * 8, stream);
    if (merr != hipSuccess) return (int)merr;
    hipLaunchKernelGGL(k_colsum_reduce, dim3(rgrid, chunks), dim3(256), 0, stream,
                       workspace, grid, (int)slab_cols, out);
    return (int)hipGetLastError();
}

// ---------------------------------------------------------------------------
// Persistent evaluation-server kernel for the gaussian linear model
// ---------------------------------------------------------------------------
//
// The per-call FIXED cost of the launch-per-eval path measured ~18 us
// (launch + wave ramp + drain) against ~7 us of actual streaming at 1e7
// rows -- the boundary dominates, which is the canonical persistent-kernel
// case.  This kernel stays resident (512 blocks = 2/CU, well under the 8
// blocks/CU the resource budget admits): block 0 polls a pinned host
// request mailbox, broadcasts {seq, theta} through device memory (sc1
// granules), every block computes its grid-stride partials, the
// last-arriver combines and publishes to the pinned result mailbox, and a
// device done-flag releases the blocks into the next iteration.
//
// EVERY spin is bounded: on exceeding its budget a block gives up and
// exits (block 0 stamps a timeout code into the result mailbox), so the
// kernel can never hang the GPU.  The host evaluator likewise times out
// and shuts the server down.

#define PK_SENTINEL 0xFFFFFFFFFFFFFFFFull
// Spin bounds. The REQUEST poll doubles as the idle lifetime: a resident
// kernel blocks any device-wide synchronize, so it exits after ~1-2 s idle
// and the host relaunches it transparently (hipStreamQuery detects exit).
#define PK_REQ_SPIN_LIMIT 2000000ll   // ~1-2 s idle at s_sleep(8) -> self-exit
#define PK_SPIN_LIMIT 800000ll        // worker bcast poll (~1-2 s)
#define PK_DONE_SPIN_LIMIT 200000ll   // completion barrier (compute is us-scale)

struct PersistentState {       // device memory
    unsigned long long bcast_seq;   // sc1-published request broadcast
    unsigned long long done_seq;    // sc1-published completion flag
    double theta[2];
};

__device__ __forceinline__ unsigned long long load_sc1_u64(const unsigned long long* p) {
    return __hip_atomic_load((const gu64_t*)p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ void store_sc1_u64(unsigned long long* p, unsigned long long v) {
    __hip_atomic_store((gu64_t*)p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

template <typename T>
__global__ __launch_bounds__(256) void k_gaussian_persistent(
    const T* __restrict__ x,
    const T* __restrict__ y,
    long long n,
    double inv_sig2,
    double logp_const,
    double* __restrict__ slab,          // [grid][3] (ws)
    unsigned* __restrict__ ticket,      // sharded tickets (ws)
    PersistentState* __restrict__ st,   // device control block
    const volatile double* __restrict__ req_host,  // pinned seqlock: [seq | a | b | seq_pre] (seq==SENTINEL -> quit)
    double* __restrict__ res_host,      // pinned: [logp ga gb | seq]
    int seqlock_on,                     // 1: one-round-trip poll; 0: detect-then-read
    int fence_mode                      // 1: __threadfence_system publish; 0: store-drain
) {
    using TR = VecTraits<T>;
    using A = typename TR::acc_t;
    constexpr int VEC = TR::VEC;
    const long long gid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    const long long gstride = (long long)gridDim.x * blockDim.x;
    const long long nvec = n / VEC;
    __shared__ double lds[4 * 3 + 4];  // reduce scratch + {seq, a, b} bcast

#define PK_STAMP(code)     if (blockIdx.x == 0 && threadIdx.x == 0) ((volatile unsigned long long*)res_host)[4] = (code);
    unsigned long long my_seq = 0;
    PK_STAMP(1)
    while (true) {
        // ---- acquire next request --------------------------------------
        if (threadIdx.x == 0) {
            unsigned long long next = my_seq + 1;
            if (blockIdx.x == 0) {
                // poll the HOST mailbox (one lane, one block).  SEQLOCK
                // layout [seq | a | b | seq_pre]: the host writes seq_pre,
                // then a, b, then seq (release), so a poll iteration whose
                // four loads (issued together -- ONE host-memory round
                // trip) sees seq >= next AND seq_pre == seq has a
                // consistent payload in hand.  Quit rides seq as
                // PK_SENTINEL.  This removes the separate post-detect
                // payload read (~1us) and the second quit-flag read the
                // round-1 protocol paid per poll.
                long long spins = 0;
                double a_req = 0.0, b_req = 0.0;
                if (seqlock_on) {
                    while (true) {
                        const unsigned long long rs =
                            ((const volatile unsigned long long*)req_host)[0];
                        const double a_r = req_host[1];
                        const double b_r = req_host[2];
                        const unsigned long long pre =
                            ((const volatile unsigned long long*)req_host)[3];
                        if (rs == PK_SENTINEL) { next = PK_SENTINEL; break; }
                        if (rs >= next && pre == rs) {
                            a_req = a_r;
                            b_req = b_r;
                            break;
                        }
                        if (rs < next)  // no new request yet: back off
                            __builtin_amdgcn_s_sleep(8);
                        // torn read (host mid-write): immediate re-poll
                        if (++spins > PK_REQ_SPIN_LIMIT) { next = PK_SENTINEL; break; }
                    }
                } else {
                    while (true) {
                        const unsigned long long rs =
                            ((const volatile unsigned long long*)req_host)[0];
                        if (rs == PK_SENTINEL) { next = PK_SENTINEL; break; }
                        if (rs >= next) break;
                        __builtin_amdgcn_s_sleep(8);
                        if (++spins > PK_REQ_SPIN_LIMIT) { next = PK_SENTINEL; break; }
                    }
                    if (next != PK_SENTINEL) {
                        a_req = req_host[1];
                        b_req = req_host[2];
                    }
                }
                if (next != PK_SENTINEL) {
                    // sc1 payload + drained sc1 flag (G16 R1: a plain store
                    // + vmcnt drain is NOT cross-XCD visible)
                    store_sc1_f64(&st->theta[0], a_req);
                    store_sc1_f64(&st->theta[1], b_req);
                    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
                    store_sc1_u64(&st->bcast_seq, next);
                    lds[12] = (double)1.0;
                    lds[13] = a_req;
                    lds[14] = b_req;
                } else {
                    store_sc1_u64(&st->bcast_seq, PK_SENTINEL);
                    lds[12] = -1.0;
                }
            } else {
                // poll the DEVICE broadcast (one lane per block)
                long long spins = 0;
                unsigned long long bs;
                while (true) {
                    bs = load_sc1_u64(&st->bcast_seq);
                    if (bs >= next || bs == PK_SENTINEL) break;
                    __builtin_amdgcn_s_sleep(16);
                    if (++spins > PK_SPIN_LIMIT) { bs = PK_SENTINEL; break; }
                }
                if (bs == PK_SENTINEL) {
                    lds[12] = -1.0;
                } else {
                    lds[12] = 1.0;
                    lds[13] = load_sc1_f64(&st->theta[0]);
                    lds[14] = load_sc1_f64(&st->theta[1]);
                }
            }
        }
        __syncthreads();
        if (lds[12] < 0.0) return;  // quit or spin give-up
        PK_STAMP(2)
        const double a = lds[13];
        const double b = lds[14];
        __syncthreads();  // lds reused by the reduction below
        my_seq += 1;

        // ---- compute this block's partials ------------------------------
        double sr, srx, sr2;
        gauss_accumulate<T>(x, y, n, gid, gstride, a, b, sr, srx, sr2);
        PK_STAMP(3)
        double acc[3] = {sr2, sr, srx};
        block_reduce_add<3>(acc, lds);
        bool last = false;
        if (threadIdx.x == 0) {
            double* s = slab + 3 * (long long)blockIdx.x;
            store_sc1_f64(&s[0], acc[0]);
            store_sc1_f64(&s[1], acc[1]);
            store_sc1_f64(&s[2], acc[2]);
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            const unsigned ngroups = gridDim.x < 8 ? gridDim.x : 8;
            const unsigned gsize = gridDim.x / ngroups;
            const unsigned grp = blockIdx.x % ngroups;
            const unsigned old = __hip_atomic_fetch_add(
                (gu32_t*)(ticket + 16 * grp), 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
            if (old % gsize == gsize - 1) {
                const unsigned t = __hip_atomic_fetch_add(
                    (gu32_t*)(ticket + 16 * 8), 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                last = (t % ngroups) == (ngroups - 1);
            }
            lds[12] = last ? 1.0 : 0.0;
        }
        __syncthreads();

        if (lds[12] != 0.0) {
            // last-arriving block: combine + publish + release the others
            double fin[3] = {0.0, 0.0, 0.0};
            for (unsigned i = threadIdx.x; i < gridDim.x; i += blockDim.x) {
#pragma unroll
                for (int k = 0; k < 3; ++k)
                    fin[k] += load_sc1_f64(&slab[3 * (long long)i + k]);
            }
            __syncthreads();
            block_reduce_add<3>(fin, lds);
            if (threadIdx.x == 0) {
                // volatile: inside the persistent loop the compiler defers /
                // elides plain mailbox stores (measured: data landed, the
                // plain u64 flag never did)
                volatile double* rh = (volatile double*)res_host;
                rh[0] = logp_const - 0.5 * inv_sig2 * fin[0];
                rh[1] = inv_sig2 * fin[1];
                rh[2] = inv_sig2 * fin[2];
                if (fence_mode) {
                    __threadfence_system();
                } else {
                    // store-drain instead of the system fence: same-box A/B
                    // measured 55.4-55.8k vs 54.7-55.7k calls/s -- ~1%,
                    // within rep noise, so the formally correct
                    // __threadfence_system stays the default and this
                    // path is kept only as the documented experiment
                    // (FED_PK_FENCE=0)
                    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
                }
                ((volatile unsigned long long*)res_host)[3] = my_seq;
                store_sc1_u64(&st->done_seq, my_seq);
            }
        }
        PK_STAMP(4)
        // ---- completion barrier (slab must not be reused early) ---------
        if (threadIdx.x == 0) {
            long long spins = 0;
            while (load_sc1_u64(&st->done_seq) < my_seq) {
                __builtin_amdgcn_s_sleep(8);
                if (++spins > PK_DONE_SPIN_LIMIT) { lds[12] = -1.0; break; }
            }
        }
        __syncthreads();
        if (lds[12] < 0.0) return;
    }
}

struct FedPersistentLinear {
    double* ws = nullptr;          // tickets + slab
    PersistentState* st = nullptr;
    double* req = nullptr;         // pinned [seq a b quit]
    double* res = nullptr;         // pinned [3 results | seq]
    hipStream_t stream = nullptr;
    unsigned long long seq = 0;
    int grid = 512;
    // launch parameters (for transparent relaunch after idle self-exit)
    const void* x = nullptr;
    const void* y = nullptr;
    long long n = 0;
    double inv_sig2 = 0;
    double logp_const = 0;
    int dtype = FED_BF16;
    void* req_dev = nullptr;
    void* res_dev = nullptr;
};

static int persistent_launch(FedPersistentLinear* e) {
    unsigned* ticket = (unsigned*)e->ws;
    double* slab = e->ws + 72;
    // Poll protocol A/B (same box, 3 reps each, r2c17): detect-then-read
    // 55.3-55.4k calls/s vs the one-iteration seqlock poll 51.3-52.0k --
    // the seqlock's four volatile host reads issue as four serialized
    // fabric transactions PER POLL, costing more than the one payload
    // read it saves.  Default is therefore the two-step poll; the
    // seqlock reader stays behind FED_PK_SEQLOCK=1 as the documented
    // negative result.
    const char* sl = getenv("FED_PK_SEQLOCK");
    const int seqlock_on = sl ? (atoi(sl) != 0) : 0;
    const char* fm = getenv("FED_PK_FENCE");
    const int fence_mode = fm ? (atoi(fm) != 0) : 1;
    switch (e->dtype) {
        case FED_BF16:
            hipLaunchKernelGGL(k_gaussian_persistent<bf16_tag>, dim3(e->grid), dim3(256), 0,
                               e->stream, (const bf16_tag*)e->x, (const bf16_tag*)e->y, e->n,
                               e->inv_sig2, e->logp_const, slab, ticket, e->st,
                               (const volatile double*)e->req_dev, (double*)e->res_dev,
                               seqlock_on, fence_mode);
            break;
        case FED_F32:
            hipLaunchKernelGGL(k_gaussian_persistent<float>, dim3(e->grid), dim3(256), 0,
                               e->stream, (const float*)e->x, (const float*)e->y, e->n,
                               e->inv_sig2, e->logp_const, slab, ticket, e->st,
                               (const volatile double*)e->req_dev, (double*)e->res_dev,
                               seqlock_on, fence_mode);
            break;
        case FED_F64:
            hipLaunchKernelGGL(k_gaussian_persistent<double>, dim3(e->grid), dim3(256), 0,
                               e->stream, (const double*)e->x, (const double*)e->y, e->n,
                               e->inv_sig2, e->logp_const, slab, ticket, e->st,
                               (const volatile double*)e->req_dev, (double*)e->res_dev,
                               seqlock_on, fence_mode);
            break;
        default:
            return -2;
    }
    return (int)hipGetLastError();
}

extern "C" {

void* fed_gaussian_persistent_start(
    const void* x, const void* y, long long n, double sigma, int dtype
) {
    FedPersistentLinear* e = new FedPersistentLinear();
    {
        const char* env = getenv("FED_PERSIST_GRID");
        if (env) {
            int g = atoi(env);
            if (g >= 8 && g <= 1024 && (g & (g - 1)) == 0) e->grid = g;
        }
    }
    const long long ws_words = 72 + 3 * (long long)e->grid;
    if (hipStreamCreateWithFlags(&e->stream, hipStreamNonBlocking) != hipSuccess ||
        hipMalloc(&e->ws, ws_words * 8) != hipSuccess ||
        hipMemset(e->ws, 0, ws_words * 8) != hipSuccess ||
        hipMalloc(&e->st, sizeof(PersistentState)) != hipSuccess ||
        hipMemset(e->st, 0, sizeof(PersistentState)) != hipSuccess ||
        hipHostMalloc((void**)&e->req, 8 * 8, hipHostMallocMapped) != hipSuccess ||
        hipHostMalloc((void**)&e->res, 8 * 8, hipHostMallocMapped) != hipSuccess) {
        delete e;
        return nullptr;
    }
    for (int i = 0; i < 8; ++i) {
        e->req[i] = 0.0;
        e->res[i] = 0.0;
    }
    void* req_dev = nullptr;
    void* res_dev = nullptr;
    if (hipHostGetDevicePointer(&req_dev, e->req, 0) != hipSuccess ||
        hipHostGetDevicePointer(&res_dev, e->res, 0) != hipSuccess) {
        delete e;
        return nullptr;
    }
    e->x = x;
    e->y = y;
    e->n = n;
    e->dtype = dtype;
    e->inv_sig2 = 1.0 / (sigma * sigma);
    e->logp_const = -0.5 * (double)n * log(2.0 * M_PI * sigma * sigma);
    e->req_dev = req_dev;
    e->res_dev = res_dev;
    if (persistent_launch(e) != 0) {
        delete e;
        return nullptr;
    }
    return e;
}

int fed_gaussian_persistent_eval(void* handle, double a, double b, double* out3) {
    FedPersistentLinear* e = (FedPersistentLinear*)handle;
    e->seq += 1;
    __atomic_store_n((unsigned long long*)&e->req[3], e->seq, __ATOMIC_RELEASE);
    e->req[1] = a;
    e->req[2] = b;
    __atomic_store_n((unsigned long long*)&e->req[0], e->seq, __ATOMIC_RELEASE);
    volatile unsigned long long* flag = ((volatile unsigned long long*)e->res) + 3;
    int relaunches = 0;
    for (long long spins = 0; spins < 4000000000LL; ++spins) {
        if (*flag >= e->seq) {
            out3[0] = e->res[0];
            out3[1] = e->res[1];
            out3[2] = e->res[2];
            return 0;
        }
        if ((spins & 0xFFFFF) == 0xFFFFF) {  // every ~1M spins (~1-2 ms)
            // server may have idle-exited (bounded request poll); relaunch --
            // the fresh kernel sees the pending req_seq and serves it
            if (hipStreamQuery(e->stream) != hipErrorNotReady) {
                if (relaunches++ > 4) return -6;
                if (persistent_launch(e) != 0) return -7;
            }
        }
    }
    return -6;
}

int fed_gaussian_persistent_debug(void* handle, double* req8, double* res8) {
    FedPersistentLinear* e = (FedPersistentLinear*)handle;
    for (int i = 0; i < 8; ++i) {
        req8[i] = e->req[i];
        res8[i] = e->res[i];
    }
    return 0;
}

int fed_gaussian_persistent_stop(void* handle) {
    FedPersistentLinear* e = (FedPersistentLinear*)handle;
    __atomic_store_n((unsigned long long*)&e->req[0], PK_SENTINEL, __ATOMIC_RELEASE);
    hipError_t err = hipStreamSynchronize(e->stream);  // kernel exits on quit
    if (e->ws) (void)hipFree(e->ws);
    if (e->st) (void)hipFree(e->st);
    if (e->req) (void)hipHostFree(e->req);
    if (e->res) (void)hipHostFree(e->res);
    if (e->stream) (void)hipStreamDestroy(e->stream);
    delete e;
    return (int)err;
}

}  // extern "C"
