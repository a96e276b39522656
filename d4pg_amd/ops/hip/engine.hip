// d4pg_amd fused D4PG learner engine — CDNA4 (gfx950 / MI355X) native.
//
// The entire D4PG train step of the reference (/root/reference/ddpg.py:200-255
// — PER sample, target forwards, C51 projection, critic CE backward, Adam,
// policy backward, Adam, target soft-update, priority write-back) runs
// on-device.  All state is device-resident: parameters in a flat slab
// (weights stored TRANSPOSED, [in][out], for coalesced forward reads),
// Adam moments, target slabs, the SoA replay store and the prioritized
// sum/min segment trees in HBM, philox4x32 RNG, and
// the beta/Adam step counters.
//
// Three execution paths, selected by batch size (Engine::enqueue_step):
//
//  * PERSISTENT (B <= 256, the flagship regime): k_step_persistent — ONE
//    kernel runs the whole step (and an nsteps loop of steps) as ~27
//    layer-parallel phases over 64 co-resident workgroups separated by a
//    software grid barrier (tree arrival + write-once go-flag).  Each
//    GEMM/dX phase tiles 4 rows x 64 cols per workgroup with register-
//    batched, double-buffered LDS staging; PER write-back and the next
//    step's tree sampling hide under the dW/Adam phases.  At B=64 the
//    step is latency-bound (PMC: 3% VALUBusy), so the design maximizes
//    memory-level parallelism, not FLOPs.  k_step_chain (D4PG_CHAIN=1) is
//    an experimental row-local chain-fused variant.
//  * MFMA (B >= 512, the wide-batch config): per-layer f32-in matrix-core
//    GEMMs (v_mfma_f32_32x32x2_f32 — exact fp32 at the full fp32 rate),
//    64x128 workgroup tiles, double-buffered LDS K-slices, split-K dW
//    with fused bias, per-level PER tree repair.
//  * per-layer VALU kernels + hipGraph capture (fallback / middle sizes,
//    incl. the row-block megakernels for 256 < B <= 512).
//
// Numerics: fp32 end-to-end, matching the reference's CPU fp32; per-thread
// dot products accumulate in ascending-k order so every path agrees with
// the eager torch oracle within fp32 tolerance.

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <cmath>
#include <vector>
#include <stdexcept>

#define HIP_CHECK(cmd) do { \
    hipError_t e_ = (cmd); \
    if (e_ != hipSuccess) { \
        char buf[256]; \
        snprintf(buf, sizeof(buf), "HIP error %s at %s:%d", \
                 hipGetErrorString(e_), __FILE__, __LINE__); \
        throw std::runtime_error(buf); \
    } \
} while (0)

namespace d4pg {

// ---------------------------------------------------------------------------
// philox4x32-10 counter-based RNG (device)
// ---------------------------------------------------------------------------
__device__ __forceinline__ uint32_t mulhi32(uint32_t a, uint32_t b) {
    return (uint32_t)(((uint64_t)a * b) >> 32);
}

struct Philox4 { uint32_t v[4]; };

__device__ inline Philox4 philox4(uint64_t seed, uint64_t ctr_hi,
                                  uint64_t ctr_lo) {
    uint32_t c0 = (uint32_t)ctr_lo, c1 = (uint32_t)(ctr_lo >> 32);
    uint32_t c2 = (uint32_t)ctr_hi, c3 = (uint32_t)(ctr_hi >> 32);
    uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
    const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
    const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
    for (int r = 0; r < 10; ++r) {
        uint32_t hi0 = mulhi32(M0, c0), lo0 = M0 * c0;
        uint32_t hi1 = mulhi32(M1, c2), lo1 = M1 * c2;
        uint32_t n0 = hi1 ^ c1 ^ k0, n1 = lo1;
        uint32_t n2 = hi0 ^ c3 ^ k1, n3 = lo0;
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
        k0 += W0; k1 += W1;
    }
    return {c0, c1, c2, c3};
}

__device__ inline float u01(uint32_t x) {           // (0,1]
    return ((float)x + 1.0f) * 2.3283064e-10f;
}

// ---------------------------------------------------------------------------
// Engine state blocks
// ---------------------------------------------------------------------------

// One linear layer's geometry inside a parameter slab.
struct LayerDesc {
    int in1, in2, out;       // in2 > 0 => concat second input
    long w_off, b_off;       // offsets into the net's slab (floats)
};

enum Act { ACT_NONE = 0, ACT_RELU = 1, ACT_TANH = 2, ACT_SOFTMAX = 3 };

// A forward job: y = act(x1 [. x2] @ Wt + b).  Grid-sliced when several
// independent jobs share a launch.
struct FwdJob {
    const float* x1; const float* x2;
    const float* wt; const float* bias;
    float* y;
    int B, in1, in2, out, act;
    int wg0, nwg_b, nwg_o;   // wg slice: blocks [wg0, wg0+nwg_b*nwg_o)
};

// Backward job for one layer: given dz (grad wrt pre-activation, [B,out]):
//   dWt[i][o] += sum_b x[b][i] dz[b][o];  db[o] += sum_b dz[b][o]
//   dx[b][i]   = sum_o dz[b][o] Wt[i][o], then * act'(h_prev)
struct BwdJob {
    const float* dz;
    const float* x1; const float* x2;       // layer inputs (concat aware)
    const float* wt;
    float* dwt; float* dbias;               // null => skip dW part
    float* dx1; float* dx2;                 // null => skip dX part(s)
    const float* h1;                        // prev activation for mask (dx1)
    int B, in1, in2, out;
    int prev_act;                           // act of PREVIOUS layer (mask)
    int wg0_dw, nwg_dw_i, nwg_dw_o;         // dW tile grid slice
    int wg0_dx, nwg_dx_b, nwg_dx_i;         // dX tile grid slice
};

struct EngineCfg {
    int obs, act, hidden, atoms, batch;
    long capacity;             // replay capacity (will be rounded to pow2 tree)
    float v_min, v_max, gamma_n, tau, lr_actor, lr_critic;
    float per_alpha, per_beta0, per_eps;
    long per_beta_iters;
    uint64_t seed;
    int is_weighting;          // apply IS weights to the CE loss
};

// device-side counters (one small block)
struct Counters {
    long long beta_t;          // PER beta schedule position (stateful)
    long long adam_t_actor;
    long long adam_t_critic;
    long long rng_epoch;       // bumped per step for fresh philox streams
    long long size;            // replay occupancy
    long long pos;             // replay ring position
    float max_priority;
    float loss_critic;         // per-step scalars (overwritten each step)
    float loss_actor;
};

// ===========================================================================
// Kernels
// ===========================================================================

// ---- tick: advance counters at the start of every step --------------------
__global__ void k_tick(Counters* c) {
    if (threadIdx.x == 0) {
        c->beta_t += 1;
        c->adam_t_actor += 1;
        c->adam_t_critic += 1;
        c->rng_epoch += 1;
        c->loss_critic = 0.f;
        c->loss_actor = 0.f;
    }
}

// ---- PER sample + gather ---------------------------------------------------
// One wave per probe: lane 0 walks the sum tree (double, exact), all lanes
// then cooperatively gather the transition row into the batch SoA.
// Also computes IS weights w = ((p/total)*N)^-beta / max_w.
__global__ void k_per_sample(
        const double* __restrict__ sum_tree, const double* __restrict__ min_tree,
        long tree_cap,
        const float* __restrict__ rs, const float* __restrict__ ra,
        const float* __restrict__ rr, const float* __restrict__ rs2,
        const float* __restrict__ rd,
        int obs, int act,
        float* __restrict__ bs, float* __restrict__ ba, float* __restrict__ br,
        float* __restrict__ bs2, float* __restrict__ bd,
        float* __restrict__ bw, long* __restrict__ bidx,
        int B, const Counters* __restrict__ cnt,
        float beta0, float beta_iters, uint64_t seed) {
    int probe = blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
    int lane = threadIdx.x & 63;
    if (probe >= B) return;
    if (probe == 0 && lane == 0) {
        // first kernel of the step: reset the per-step loss accumulators
        // (counter increments happen in k_per_update at step end)
        const_cast<Counters*>(cnt)->loss_critic = 0.f;
        const_cast<Counters*>(cnt)->loss_actor = 0.f;
    }

    long long n = cnt->size;
    double total = sum_tree[1];
    float frac = fminf((float)((double)cnt->beta_t / beta_iters), 1.0f);
    float beta = beta0 + frac * (1.0f - beta0);

    long idx;
    if (lane == 0) {
        Philox4 r = philox4(seed, (uint64_t)cnt->rng_epoch, (uint64_t)probe);
        double mass = (double)u01(r.v[0]) * total;
        long node = 1;
        // 4-ary descent over the binary layout: a stored parent is
        // bit-exactly the double-sum of its children, so comparing
        // against grandchildren sums reproduces two binary steps with
        // ONE contiguous 32 B load — half the dependent-load chain
        // (20 levels -> 10 loads at 1e6 leaves).
        while (node < tree_cap / 2) {
            long gb = 4 * node;
            double g0 = sum_tree[gb], g1 = sum_tree[gb + 1],
                   g2 = sum_tree[gb + 2];
            double s01 = g0 + g1;
            if (mass <= s01) {
                if (mass <= g0) node = gb;
                else { mass -= g0; node = gb + 1; }
            } else {
                mass -= s01;
                if (mass <= g2) node = gb + 2;
                else { mass -= g2; node = gb + 3; }
            }
        }
        if (node < tree_cap) {                 // odd final level
            double ls = sum_tree[2 * node];
            if (mass > ls) { mass -= ls; node = 2 * node + 1; }
            else           { node = 2 * node; }
        }
        idx = node - tree_cap;
        if (idx >= n) idx = n - 1;          // fp-roundoff guard at range top
        bidx[probe] = idx;
        // IS weight
        double p = sum_tree[tree_cap + idx] / total;
        double p_min = min_tree[1] / total;
        double max_w = pow(p_min * (double)n, (double)-beta);
        bw[probe] = (float)(pow(p * (double)n, (double)-beta) / max_w);
        br[probe] = rr[idx];
        bd[probe] = rd[idx];
    }
    idx = __shfl(idx, 0, 64);
    for (int k = lane; k < obs; k += 64) {
        bs[(long)probe * obs + k] = rs[idx * obs + k];
        bs2[(long)probe * obs + k] = rs2[idx * obs + k];
    }
    for (int k = lane; k < act; k += 64)
        ba[(long)probe * act + k] = ra[idx * act + k];
}

// ---- generic fused forward -------------------------------------------------
// Tile: TB rows x TO cols per workgroup; 256 threads; LDS-staged x and Wt
// chunks.  ACT_SOFTMAX requires out <= 64 and uses one wave per row.
#define TB 4
#define TO 64
// max fan-in a fwd workgroup stages (hidden(1024) + act margin)
#define FWD_XMAX 1032

__device__ inline void fwd_one(const FwdJob& j, int wg) {
    int rel = wg - j.wg0;
    int bt = rel / j.nwg_o;            // which row tile
    int ot = rel % j.nwg_o;            // which col tile
    int b0 = bt * TB, o0 = ot * TO;
    int in_total = j.in1 + j.in2;

    // x rows staged ONCE (LDS broadcast reads after); weights read straight
    // from global — coalesced 256 B per wave per k, pipelined by the unroll
    // (the previous sync-staged-chunks version was latency-bound at ~20 us
    // per launch; this form measures ~3-5 us at B=64, H=256).
    __shared__ float xs[TB][FWD_XMAX];

    int tid = threadIdx.x;
    for (int t = tid; t < TB * in_total; t += 256) {
        int bb = t / in_total, kk = t % in_total;
        int gb = b0 + bb;
        float v = 0.f;
        if (gb < j.B)
            v = (kk < j.in1) ? j.x1[(long)gb * j.in1 + kk]
                             : j.x2[(long)gb * j.in2 + (kk - j.in1)];
        xs[bb][kk] = v;
    }
    __syncthreads();

    int tb = tid / TO;                 // 0..TB-1
    int to = tid % TO;                 // 0..TO-1
    int b = b0 + tb, o = o0 + to;

    float acc = 0.f;
    if (o < j.out) {
        const float* wcol = j.wt + o;
#pragma unroll 8
        for (int k = 0; k < in_total; ++k)
            acc += xs[tb][k] * wcol[(long)k * j.out];
    }

    if (b < j.B && o < j.out) {
        acc += j.bias[o];
        if (j.act == ACT_RELU) acc = fmaxf(acc, 0.f);
        else if (j.act == ACT_TANH) acc = tanhf(acc);
    }
    if (j.act == ACT_SOFTMAX) {
        // one wave handles one row (TO==64 lanes over out<=64 columns)
        float v = (b < j.B && o < j.out) ? acc : -INFINITY;
        float mx = v;
        for (int s = 32; s > 0; s >>= 1) mx = fmaxf(mx, __shfl_xor(mx, s, 64));
        float e = (b < j.B && o < j.out) ? __expf(v - mx) : 0.f;
        float sum = e;
        for (int s = 32; s > 0; s >>= 1) sum += __shfl_xor(sum, s, 64);
        if (b < j.B && o < j.out) j.y[(long)b * j.out + o] = e / sum;
    } else if (b < j.B && o < j.out) {
        j.y[(long)b * j.out + o] = acc;
    }
}

__global__ void k_fwd3(FwdJob j0, FwdJob j1, FwdJob j2, int njobs) {
    int wg = blockIdx.x;
    if (njobs > 0 && wg >= j0.wg0 && wg < j0.wg0 + j0.nwg_b * j0.nwg_o)
        { fwd_one(j0, wg); return; }
    if (njobs > 1 && wg >= j1.wg0 && wg < j1.wg0 + j1.nwg_b * j1.nwg_o)
        { fwd_one(j1, wg); return; }
    if (njobs > 2 && wg >= j2.wg0 && wg < j2.wg0 + j2.nwg_b * j2.nwg_o)
        { fwd_one(j2, wg); return; }
}

// ---- C51 categorical projection (K3) ---------------------------------------
// One wave per batch row; lane j < K owns atom j.  Mass split accumulated in
// an LDS row via LDS atomics (no global scatter races).  Equal-bin handling
// identical to algo/projection.py (index-adjust so weights become (0,1)).
__global__ void k_project(const float* __restrict__ p_t,
                          const float* __restrict__ r,
                          const float* __restrict__ d,
                          float* __restrict__ m,
                          int B, int K, float v_min, float v_max,
                          float gamma_n) {
    int row = blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
    int lane = threadIdx.x & 63;
    int wrow = threadIdx.x / 64;
    extern __shared__ float lm[];                 // [waves][K]
    float* mrow = lm + wrow * K;
    for (int k = lane; k < K; k += 64) mrow[k] = 0.f;
    __builtin_amdgcn_wave_barrier();
    if (row < B && lane < K) {
        float delta = (v_max - v_min) / (K - 1);
        float z = v_min + lane * delta;
        float tz = r[row] + gamma_n * (1.f - d[row]) * z;
        tz = fminf(v_max, fmaxf(v_min, tz));
        float b = (tz - v_min) / delta;
        int l = (int)floorf(b);
        int u = (int)ceilf(b);
        if (l == u) { if (u > 0) l -= 1; else u += 1; }
        float p = p_t[(long)row * K + lane];
        atomicAdd(&mrow[l], p * ((float)u - b));
        atomicAdd(&mrow[u], p * (b - (float)l));
    }
    __builtin_amdgcn_wave_barrier();
    if (row < B)
        for (int k = lane; k < K; k += 64)
            m[(long)row * K + k] = mrow[k];
}

// ---- critic CE gradient + PER priorities (K4+K5) ---------------------------
// dlogits = (q - m) * scale / B   (softmax+CE fused gradient)
// priority = |-(sum m*q)| + eps = sum(m*q) + eps (reference proxy,
// ddpg.py:220-222).  scale = IS weight when is_weighting.
__global__ void k_ce_grad(const float* __restrict__ q,
                          const float* __restrict__ m,
                          const float* __restrict__ w,
                          float* __restrict__ dlogits,
                          float* __restrict__ pri,
                          Counters* cnt,
                          int B, int K, float per_eps, int is_weighting) {
    int row = blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
    int lane = threadIdx.x & 63;
    if (row >= B) return;
    float qv = 0.f, mv = 0.f;
    if (lane < K) {
        qv = q[(long)row * K + lane];
        mv = m[(long)row * K + lane];
    }
    float dot = mv * qv;                    // for priority
    float ce = -mv * __logf(qv + 1e-10f);   // for loss scalar
    for (int s = 32; s > 0; s >>= 1) {
        dot += __shfl_xor(dot, s, 64);
        ce  += __shfl_xor(ce, s, 64);
    }
    float scale = (is_weighting && w) ? w[row] : 1.f;
    if (lane < K)
        dlogits[(long)row * K + lane] = scale * (qv - mv) / (float)B;
    if (lane == 0) {
        pri[row] = dot + per_eps;
        atomicAdd(&cnt->loss_critic, scale * ce / (float)B);
    }
}

// ---- row softmax (wide path: MFMA computes logits, this normalizes) --------
__global__ void k_row_softmax(const float* __restrict__ logits,
                              float* __restrict__ y, int B, int K) {
    int row = blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
    int lane = threadIdx.x & 63;
    if (row >= B) return;
    float v = (lane < K) ? logits[(long)row * K + lane] : -INFINITY;
    float mx = v;
    for (int s = 32; s > 0; s >>= 1) mx = fmaxf(mx, __shfl_xor(mx, s, 64));
    float e = (lane < K) ? __expf(v - mx) : 0.f;
    float sum = e;
    for (int s = 32; s > 0; s >>= 1) sum += __shfl_xor(sum, s, 64);
    if (lane < K) y[(long)row * K + lane] = e / sum;
}

// ---- fused C51 projection + CE grad + priorities (wide path) ----------------
// (the persistent path fuses these in p_project_ce; this is the standalone
// twin for the per-layer/MFMA path — saves a launch + the m round-trip)
__global__ void k_project_ce(const float* __restrict__ p_t,
                             const float* __restrict__ r,
                             const float* __restrict__ d,
                             const float* __restrict__ q,
                             const float* __restrict__ w,
                             float* __restrict__ m_out,
                             float* __restrict__ dlogits,
                             float* __restrict__ pri, Counters* cnt,
                             int B, int K, float v_min, float v_max,
                             float gamma_n, float per_eps,
                             int is_weighting) {
    int lane = threadIdx.x & 63;
    int wrow = threadIdx.x / 64;
    int wpb = blockDim.x / 64;
    extern __shared__ float lm[];
    float* mrow = lm + wrow * 64;
    __shared__ float loss_red[8];
    float ce_acc = 0.f;
    // grid-stride over rows with a single per-wg loss atomic at the end:
    // one atomicAdd per ROW to the shared loss scalar serialized ~50 us
    // at B=4096 (4096 same-address RMWs)
    for (int row = blockIdx.x * wpb + wrow; row < B;
         row += gridDim.x * wpb) {
    mrow[lane] = 0.f;
    __builtin_amdgcn_wave_barrier();
    if (row < B && lane < K) {
        float delta = (v_max - v_min) / (K - 1);
        float z = v_min + lane * delta;
        float tz = r[row] + gamma_n * (1.f - d[row]) * z;
        tz = fminf(v_max, fmaxf(v_min, tz));
        float b = (tz - v_min) / delta;
        int l = (int)floorf(b), u = (int)ceilf(b);
        if (l == u) { if (u > 0) l -= 1; else u += 1; }
        float p = p_t[(long)row * K + lane];
        atomicAdd(&mrow[l], p * ((float)u - b));
        atomicAdd(&mrow[u], p * (b - (float)l));
    }
    __builtin_amdgcn_wave_barrier();
    float mv = (lane < K) ? mrow[lane] : 0.f;
    float qv = (lane < K) ? q[(long)row * K + lane] : 0.f;
    if (lane < K) m_out[(long)row * K + lane] = mv;
    float dot = mv * qv;
    float ce = -mv * __logf(qv + 1e-10f);
    for (int ss = 32; ss > 0; ss >>= 1) {
        dot += __shfl_xor(dot, ss, 64);
        ce += __shfl_xor(ce, ss, 64);
    }
    float scale = (is_weighting && w) ? w[row] : 1.f;
    if (lane < K)
        dlogits[(long)row * K + lane] = scale * (qv - mv) / (float)B;
    if (lane == 0) {
        pri[row] = dot + per_eps;
        ce_acc += scale * ce / (float)B;
    }
    __builtin_amdgcn_wave_barrier();
    }
    __syncthreads();
    if (lane == 0) loss_red[wrow] = ce_acc;
    __syncthreads();
    if (threadIdx.x == 0) {
        float s = 0.f;
        for (int wv = 0; wv < wpb; ++wv) s += loss_red[wv];
        atomicAdd(&cnt->loss_critic, s);
    }
}

// ---- policy gradient through the softmax head (K6 seed) ---------------------
// L = -(1/B) sum_b sum_k q_k z_k  =>  dlogits_j = -q_j (z_j - E_q[z]) / B
__global__ void k_policy_grad(const float* __restrict__ q,
                              float* __restrict__ dlogits,
                              Counters* cnt,
                              int B, int K, float v_min, float v_max) {
    int lane = threadIdx.x & 63;
    int wrow = threadIdx.x / 64;
    int wpb = blockDim.x / 64;
    __shared__ float loss_red[8];
    float loss_acc = 0.f;
    float delta = (v_max - v_min) / (K - 1);
    float z = v_min + lane * delta;
    // grid-stride rows + one per-wg loss atomic (see k_project_ce)
    for (int row = blockIdx.x * wpb + wrow; row < B;
         row += gridDim.x * wpb) {
        float qv = (lane < K) ? q[(long)row * K + lane] : 0.f;
        float e = qv * z;
        for (int s = 32; s > 0; s >>= 1) e += __shfl_xor(e, s, 64);
        if (lane < K)
            dlogits[(long)row * K + lane] = -qv * (z - e) / (float)B;
        if (lane == 0) loss_acc += -e / (float)B;
    }
    __syncthreads();
    if (lane == 0) loss_red[wrow] = loss_acc;
    __syncthreads();
    if (threadIdx.x == 0) {
        float s = 0.f;
        for (int wv = 0; wv < wpb; ++wv) s += loss_red[wv];
        atomicAdd(&cnt->loss_actor, s);
    }
}

__device__ inline float act_mask(int act, float h) {
    if (act == ACT_RELU) return h > 0.f ? 1.f : 0.f;
    if (act == ACT_TANH) return 1.f - h * h;
    return 1.f;
}

// ===========================================================================
// Row-block megakernels (small-batch learner path)
// ===========================================================================
// Key observation: until the dW reduction, every quantity in the D4PG train
// step is per-batch-row: forwards, softmax, the C51 projection, CE/policy
// gradients and the whole backward-dX chain.  So ONE WAVE owns ONE row and
// runs the entire chain against L2-resident weights with zero barriers —
// the step then needs only 8 kernels (sample, critic-row-block, critic-dW,
// critic-Adam+lerp, policy-row-block, actor-dW, actor-Adam+lerp, PER
// write-back), which matters because at B=64 the step is bounded by the
// ~1.5-5 us per-kernel floor, not by FLOPs.

// per-wave LDS carve (floats): [xb XMAX][yb XMAX][h1 H][h2 H][h3 H]
// [aa 32][row0 64][row1 64][row2 64]
struct RowLds {
    float *xb, *yb, *h1, *h2, *h3, *aa, *r0, *r1, *r2;
};

__device__ inline RowLds rb_carve(float* base, int wid, int H) {
    long per = 2 * FWD_XMAX + 3 * H + 32 + 3 * 64;
    float* p = base + (long)wid * per;
    RowLds L;
    L.xb = p;               p += FWD_XMAX;
    L.yb = p;               p += FWD_XMAX;
    L.h1 = p;               p += H;
    L.h2 = p;               p += H;
    L.h3 = p;               p += H;
    L.aa = p;               p += 32;
    L.r0 = p;               p += 64;
    L.r1 = p;               p += 64;
    L.r2 = p;
    return L;
}

__host__ __device__ inline long rb_lds_bytes(int H) {
    return 4L * (2 * FWD_XMAX + 3 * H + 32 + 3 * 64) * 4;
}

// y[o] = act(sum_k xb[k] * Wt[k][o] + b[o]), lane-parallel over o.
__device__ inline void rb_fwd(const float* xb, float* yb, const float* wt,
                              const float* bias, int in_total, int out,
                              int act_kind, float* grow, int lane) {
    for (int oc = 0; oc < out; oc += 64) {
        int o = oc + lane;
        float acc = 0.f;
        if (o < out) {
            const float* wcol = wt + o;
#pragma unroll 16
            for (int k = 0; k < in_total; ++k)
                acc += xb[k] * wcol[(long)k * out];
            acc += bias[o];
            if (act_kind == ACT_RELU) acc = fmaxf(acc, 0.f);
            else if (act_kind == ACT_TANH) acc = tanhf(acc);
            yb[o] = acc;
            if (grow) grow[o] = acc;
        }
    }
}

// softmax head (out <= 64): q[o] = softmax(logits)[o]
__device__ inline void rb_fwd_softmax(const float* xb, float* qrow,
                                      const float* wt, const float* bias,
                                      int in_total, int out, float* grow,
                                      int lane) {
    float acc = -INFINITY;
    if (lane < out) {
        const float* wcol = wt + lane;
        float a = 0.f;
#pragma unroll 16
        for (int k = 0; k < in_total; ++k)
            a += xb[k] * wcol[(long)k * out];
        acc = a + bias[lane];
    }
    float mx = acc;
    for (int s = 32; s > 0; s >>= 1) mx = fmaxf(mx, __shfl_xor(mx, s, 64));
    float e = (lane < out) ? __expf(acc - mx) : 0.f;
    float sum = e;
    for (int s = 32; s > 0; s >>= 1) sum += __shfl_xor(sum, s, 64);
    if (lane < out) {
        float q = e / sum;
        qrow[lane] = q;
        if (grow) grow[lane] = q;
    }
}

// dx[i] = (sum_o dz[o] * Wt[i][o]) * act'(hprev[i]), lane-parallel over i.
__device__ inline void rb_bwd_dx(const float* dzb, const float* wt,
                                 int in_lo, int in_hi, int in_total, int out,
                                 const float* hprev, int prev_act,
                                 float* dxb, float* grow, int lane) {
    for (int ic = in_lo; ic < in_hi; ic += 64) {
        int i = ic + lane;
        float acc = 0.f;
        if (i < in_hi) {
            const float* wrow = wt + (long)i * out;
#pragma unroll 16
            for (int o = 0; o < out; ++o)
                acc += dzb[o] * wrow[o];
            if (hprev)
                acc *= act_mask(prev_act, hprev[i - in_lo]);
            if (dxb) dxb[i - in_lo] = acc;
            if (grow) grow[i - in_lo] = acc;
        }
    }
}

// Layer pointer bundle for a net slab.
struct NetPtrs { const float *w1, *b1, *w2, *b2, *w3, *b3, *w4, *b4; };

__device__ inline NetPtrs net_ptrs(const float* slab, const LayerDesc* l) {
    return {slab + l[0].w_off, slab + l[0].b_off,
            slab + l[1].w_off, slab + l[1].b_off,
            slab + l[2].w_off, slab + l[2].b_off,
            slab + l[3].w_off, slab + l[3].b_off};
}

// All geometry/pointers for the two row-block kernels.
struct RowBlockArgs {
    // dims
    int B, O, A, H, K;
    float v_min, v_max, gamma_n, per_eps;
    int is_weighting;
    // slabs
    const float *p_actor, *p_actor_t, *p_critic, *p_critic_t;
    LayerDesc al[4], cl[4];
    // batch
    const float *bs, *ba, *br, *bs2, *bd, *bw;
    // outputs / saved activations
    float *a2, *p_t, *m_proj, *q, *dlog, *pri;
    float *c_h1, *c_h2, *c_h3, *d1, *d2, *d3;
    float *pa_h1, *pa_h2, *pa_h3, *a_out;
    float *pc_h1, *pc_h2, *pc_h3, *pq;
    float *az1, *az2, *az3, *adz;
    Counters* cnt;
};

// K2: target forwards + projection + critic forward + CE grad + priorities
// + backward-dX chain.  One wave per row, no barriers.
__global__ void __launch_bounds__(256)
k_critic_rowblock(RowBlockArgs g) {
    extern __shared__ __attribute__((aligned(16))) float smem[];
    int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
    int b = blockIdx.x * 4 + wid;
    if (b >= g.B) return;
    RowLds L = rb_carve(smem, wid, g.H);
    NetPtrs at = net_ptrs(g.p_actor_t, g.al);
    NetPtrs ct = net_ptrs(g.p_critic_t, g.cl);
    NetPtrs c = net_ptrs(g.p_critic, g.cl);
    int O = g.O, A = g.A, H = g.H, K = g.K;

    // ---- actor_target(s2) -> a2 ----
    for (int k = lane; k < O; k += 64) L.xb[k] = g.bs2[(long)b * O + k];
    rb_fwd(L.xb, L.yb, at.w1, at.b1, O, H, ACT_RELU, nullptr, lane);
    rb_fwd(L.yb, L.xb, at.w2, at.b2, H, H, ACT_NONE, nullptr, lane);
    rb_fwd(L.xb, L.yb, at.w3, at.b3, H, H, ACT_RELU, nullptr, lane);
    rb_fwd(L.yb, L.aa, at.w4, at.b4, H, A, ACT_TANH,
           g.a2 + (long)b * A, lane);

    // ---- critic_target(s2, a2) -> p_t ----
    for (int k = lane; k < O; k += 64) L.xb[k] = g.bs2[(long)b * O + k];
    rb_fwd(L.xb, L.yb, ct.w1, ct.b1, O, H, ACT_RELU, nullptr, lane);
    for (int k = lane; k < A; k += 64) L.yb[H + k] = L.aa[k];
    rb_fwd(L.yb, L.xb, ct.w2, ct.b2, H + A, H, ACT_RELU, nullptr, lane);
    rb_fwd(L.xb, L.yb, ct.w3, ct.b3, H, H, ACT_RELU, nullptr, lane);
    rb_fwd_softmax(L.yb, L.r0, ct.w4, ct.b4, H, K,
                   g.p_t + (long)b * K, lane);          // r0 = p_t row

    // ---- C51 projection of r0 -> r1 (m row) ----
    {
        float delta = (g.v_max - g.v_min) / (K - 1);
        float rr = g.br[b], dd = g.bd[b];
        for (int k = lane; k < K; k += 64) L.r1[k] = 0.f;
        if (lane < K) {
            float z = g.v_min + lane * delta;
            float tz = rr + g.gamma_n * (1.f - dd) * z;
            tz = fminf(g.v_max, fmaxf(g.v_min, tz));
            float bj = (tz - g.v_min) / delta;
            int l = (int)floorf(bj), u = (int)ceilf(bj);
            if (l == u) { if (u > 0) l -= 1; else u += 1; }
            float p = L.r0[lane];
            atomicAdd(&L.r1[l], p * ((float)u - bj));
            atomicAdd(&L.r1[u], p * (bj - (float)l));
        }
        for (int k = lane; k < K; k += 64)
            g.m_proj[(long)b * K + k] = L.r1[k];
    }

    // ---- critic(s, a) -> q (saving h1..h3) ----
    for (int k = lane; k < O; k += 64) L.xb[k] = g.bs[(long)b * O + k];
    rb_fwd(L.xb, L.h1, c.w1, c.b1, O, H, ACT_RELU,
           g.c_h1 + (long)b * H, lane);
    for (int k = lane; k < H; k += 64) L.xb[k] = L.h1[k];
    for (int k = lane; k < A; k += 64) L.xb[H + k] = g.ba[(long)b * A + k];
    rb_fwd(L.xb, L.h2, c.w2, c.b2, H + A, H, ACT_RELU,
           g.c_h2 + (long)b * H, lane);
    rb_fwd(L.h2, L.h3, c.w3, c.b3, H, H, ACT_RELU,
           g.c_h3 + (long)b * H, lane);
    rb_fwd_softmax(L.h3, L.r0, c.w4, c.b4, H, K, g.q + (long)b * K, lane);

    // ---- CE grad + priority (r0 = q, r1 = m) ----
    {
        float qv = (lane < K) ? L.r0[lane] : 0.f;
        float mv = (lane < K) ? L.r1[lane] : 0.f;
        float dot = mv * qv;
        float ce = -mv * __logf(qv + 1e-10f);
        for (int s = 32; s > 0; s >>= 1) {
            dot += __shfl_xor(dot, s, 64);
            ce += __shfl_xor(ce, s, 64);
        }
        float scale = (g.is_weighting && g.bw) ? g.bw[b] : 1.f;
        if (lane < K) {
            float dz = scale * (qv - mv) / (float)g.B;
            L.r2[lane] = dz;
            g.dlog[(long)b * K + lane] = dz;
        }
        if (lane == 0) {
            g.pri[b] = dot + g.per_eps;
            atomicAdd(&g.cnt->loss_critic, scale * ce / (float)g.B);
        }
    }

    // ---- backward-dX chain (r2 = dlogits) ----
    rb_bwd_dx(L.r2, c.w4, 0, H, H, K, L.h3, ACT_RELU, L.xb,
              g.d3 + (long)b * H, lane);                 // dz3
    rb_bwd_dx(L.xb, c.w3, 0, H, H, H, L.h2, ACT_RELU, L.yb,
              g.d2 + (long)b * H, lane);                 // dz2
    rb_bwd_dx(L.yb, c.w2, 0, H, H + A, H, L.h1, ACT_RELU, L.xb,
              g.d1 + (long)b * H, lane);                 // dz1 (h-part only)
}

// K5: actor forward + critic(s, actor(s)) + policy grad + dX chain down to
// the actor's per-layer dz's.  One wave per row.
__global__ void __launch_bounds__(256)
k_policy_rowblock(RowBlockArgs g) {
    extern __shared__ __attribute__((aligned(16))) float smem[];
    int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
    int b = blockIdx.x * 4 + wid;
    if (b >= g.B) return;
    RowLds L = rb_carve(smem, wid, g.H);
    NetPtrs a = net_ptrs(g.p_actor, g.al);
    NetPtrs c = net_ptrs(g.p_critic, g.cl);
    int O = g.O, A = g.A, H = g.H, K = g.K;

    // ---- actor(s): save pa_h1..3 in h1..h3 ----
    for (int k = lane; k < O; k += 64) L.xb[k] = g.bs[(long)b * O + k];
    rb_fwd(L.xb, L.h1, a.w1, a.b1, O, H, ACT_RELU,
           g.pa_h1 + (long)b * H, lane);
    rb_fwd(L.h1, L.h2, a.w2, a.b2, H, H, ACT_NONE,
           g.pa_h2 + (long)b * H, lane);
    rb_fwd(L.h2, L.h3, a.w3, a.b3, H, H, ACT_RELU,
           g.pa_h3 + (long)b * H, lane);
    rb_fwd(L.h3, L.aa, a.w4, a.b4, H, A, ACT_TANH,
           g.a_out + (long)b * A, lane);

    // ---- critic(s, a_out): keep pc_h1..3 in xb/yb + globals ----
    for (int k = lane; k < O; k += 64) L.xb[k] = g.bs[(long)b * O + k];
    rb_fwd(L.xb, L.yb, c.w1, c.b1, O, H, ACT_RELU,
           g.pc_h1 + (long)b * H, lane);
    for (int k = lane; k < A; k += 64) L.yb[H + k] = L.aa[k];
    rb_fwd(L.yb, L.xb, c.w2, c.b2, H + A, H, ACT_RELU,
           g.pc_h2 + (long)b * H, lane);
    rb_fwd(L.xb, L.yb, c.w3, c.b3, H, H, ACT_RELU,
           g.pc_h3 + (long)b * H, lane);
    // yb holds pc_h3; xb holds pc_h2 — careful reuse below.
    rb_fwd_softmax(L.yb, L.r0, c.w4, c.b4, H, K, g.pq + (long)b * K, lane);

    // ---- policy head gradient: pd4 = -q (z - E[z]) / B ----
    {
        float delta = (g.v_max - g.v_min) / (K - 1);
        float z = g.v_min + lane * delta;
        float qv = (lane < K) ? L.r0[lane] : 0.f;
        float e = qv * z;
        for (int s = 32; s > 0; s >>= 1) e += __shfl_xor(e, s, 64);
        if (lane < K) L.r2[lane] = -qv * (z - e) / (float)g.B;
        if (lane == 0) atomicAdd(&g.cnt->loss_actor, -e / (float)g.B);
    }

    // ---- critic dX (no dW): r2 -> da -> actor dz chain ----
    // masks for the critic activations come from the global rows just
    // written (pc_h3/pc_h2); the actor's h1..h3 are still LDS-resident.
    const float* gpc_h3 = g.pc_h3 + (long)b * H;
    const float* gpc_h2 = g.pc_h2 + (long)b * H;
    // dz3' = (r2 @ W4c^T) * relu'(pc_h3)           -> xb
    rb_bwd_dx(L.r2, c.w4, 0, H, H, K, gpc_h3, ACT_RELU, L.xb, nullptr, lane);
    // dz2' = (xb @ W3c^T) * relu'(pc_h2)           -> yb
    rb_bwd_dx(L.xb, c.w3, 0, H, H, H, gpc_h2, ACT_RELU, L.yb, nullptr, lane);
    // da = action slice of (yb @ W2c^T)            -> r1[0..A)
    rb_bwd_dx(L.yb, c.w2, H, H + A, H + A, H, nullptr, ACT_NONE, L.r1,
              nullptr, lane);
    // adz = da * (1 - a_out^2)                     -> r1, stored
    if (lane < A) {
        float y = L.aa[lane];
        float v = L.r1[lane] * (1.f - y * y);
        L.r1[lane] = v;
        g.adz[(long)b * A + lane] = v;
    }
    // actor dz chain: az3 = (r1 @ W4a^T)*relu'(h3) -> xb
    rb_bwd_dx(L.r1, a.w4, 0, H, H, A, L.h3, ACT_RELU, L.xb,
              g.az3 + (long)b * H, lane);
    // az2 = (xb @ W3a^T) * 1 (h2 had no act)       -> yb
    rb_bwd_dx(L.xb, a.w3, 0, H, H, H, nullptr, ACT_NONE, L.yb,
              g.az2 + (long)b * H, lane);
    // az1 = (yb @ W2a^T) * relu'(h1)               -> xb
    rb_bwd_dx(L.yb, a.w2, 0, H, H, H, L.h1, ACT_RELU, L.xb,
              g.az1 + (long)b * H, lane);
}

// ---- generic fused backward (dW + db + dX in one launch) --------------------
#define BWT 16          // dW tile: BWT_i x BWT_o, thread per (i,o)
#define BXB 8           // dX tile rows (8 * FWD_XMAX floats of dz staged)
#define BXI 32          // dX tile cols (BXB * BXI == 256 threads)


__device__ inline void bwd_one(const BwdJob& j, int wg) {
    int in_total = j.in1 + j.in2;
    int tid = threadIdx.x;

    if (j.dwt && wg >= j.wg0_dw && wg < j.wg0_dw + j.nwg_dw_i * j.nwg_dw_o) {
        // ---- dW part: tile [BWT i x BWT o]; x^T and dz slices staged per
        // 64-row batch chunk, then pure-LDS MACs (one chunk at B=64).
        int rel = wg - j.wg0_dw;
        int it = rel / j.nwg_dw_o, ot = rel % j.nwg_dw_o;
        int i0 = it * BWT, o0 = ot * BWT;
        __shared__ float xs[64][BWT + 1];            // [bchunk][i]
        __shared__ float zs[64][BWT + 1];            // [bchunk][o]
        int ti = tid / BWT, to = tid % BWT;          // thread -> (i, o)
        float acc = 0.f, accb = 0.f;
        for (int bc = 0; bc < j.B; bc += 64) {
            for (int t = tid; t < 64 * BWT; t += 256) {
                int bb = t / BWT, ii = t % BWT;
                int gb = bc + bb, gi = i0 + ii;
                float xv = 0.f;
                if (gb < j.B && gi < in_total)
                    xv = (gi < j.in1) ? j.x1[(long)gb * j.in1 + gi]
                                      : j.x2[(long)gb * j.in2 + (gi - j.in1)];
                xs[bb][ii] = xv;
                int oo = ii, go = o0 + oo;
                zs[bb][oo] = (gb < j.B && go < j.out)
                    ? j.dz[(long)gb * j.out + go] : 0.f;
            }
            __syncthreads();
#pragma unroll 16
            for (int bb = 0; bb < 64; ++bb) {
                acc += xs[bb][ti] * zs[bb][to];
                if (ti == 0) accb += zs[bb][to];
            }
            __syncthreads();
        }
        int gi = i0 + ti, go = o0 + to;
        if (gi < in_total && go < j.out)
            j.dwt[(long)gi * j.out + go] = acc;
        if (ti == 0 && go < j.out && i0 == 0 && j.dbias)
            j.dbias[go] = accb;
        return;
    }

    if (wg >= j.wg0_dx && wg < j.wg0_dx + j.nwg_dx_b * j.nwg_dx_i) {
        // ---- dX part: tile [BXB b x BXI i]; dz rows staged once, Wt rows
        // read straight from global (per-thread sequential stream, L1/L2
        // cached; same latency-pipelining rationale as fwd_one).
        int rel = wg - j.wg0_dx;
        int bt = rel / j.nwg_dx_i, it = rel % j.nwg_dx_i;
        int b0 = bt * BXB, i0 = it * BXI;
        __shared__ float zsx[BXB][FWD_XMAX];   // 8 rows x out<=1032 = 33 KB
        int tb = tid / BXI, ti = tid % BXI;
        int b = b0 + tb, i = i0 + ti;
        for (int t = tid; t < BXB * j.out; t += 256) {
            int bb = t / j.out, oo = t % j.out;
            int gb = b0 + bb;
            zsx[bb][oo] = (gb < j.B)
                ? j.dz[(long)gb * j.out + oo] : 0.f;
        }
        __syncthreads();
        float acc = 0.f;
        if (i < in_total) {
            const float* wrow = j.wt + (long)i * j.out;
#pragma unroll 8
            for (int oo = 0; oo < j.out; ++oo)
                acc += zsx[tb][oo] * wrow[oo];
        }
        if (b < j.B && i < in_total) {
            if (i < j.in1) {
                if (j.dx1) {
                    float h = j.h1 ? j.h1[(long)b * j.in1 + i] : 0.f;
                    j.dx1[(long)b * j.in1 + i] =
                        acc * act_mask(j.prev_act, h);
                }
            } else if (j.dx2) {
                j.dx2[(long)b * j.in2 + (i - j.in1)] = acc;  // raw (concat 2nd)
            }
        }
    }
}

__global__ void k_bwd(BwdJob j) { bwd_one(j, blockIdx.x); }

// four dW-only jobs batched into one launch (row-block path: all layer
// weight gradients of one net in a single kernel)
__global__ void k_bwd4(BwdJob j0, BwdJob j1, BwdJob j2, BwdJob j3) {
    int wg = blockIdx.x;
    if (wg < j1.wg0_dw) { bwd_one(j0, wg); return; }
    if (wg < j2.wg0_dw) { bwd_one(j1, wg); return; }
    if (wg < j3.wg0_dw) { bwd_one(j2, wg); return; }
    bwd_one(j3, wg);
}

// tanh backward at the actor output: dz4 = da * (1 - a_out^2)
__global__ void k_tanh_bwd(const float* __restrict__ da,
                           const float* __restrict__ a_out,
                           float* __restrict__ dz, long n) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) {
        float y = a_out[i];
        dz[i] = da[i] * (1.f - y * y);
    }
}

// ---- fused Adam over a flat slab (K8) --------------------------------------
__global__ void k_adam(float* __restrict__ p, const float* __restrict__ g,
                       float* __restrict__ m, float* __restrict__ v,
                       long n, float lr, float b1, float b2, float eps,
                       const Counters* cnt, int is_actor) {
    long long t = is_actor ? cnt->adam_t_actor : cnt->adam_t_critic;
    float bc1 = 1.f - __powf(b1, (float)t);
    float bc2 = 1.f - __powf(b2, (float)t);
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        float gi = g[i];
        float mi = b1 * m[i] + (1.f - b1) * gi;
        float vi = b2 * v[i] + (1.f - b2) * gi * gi;
        m[i] = mi; v[i] = vi;
        p[i] -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
    }
}

// ---- fused Adam + target soft-update in one pass (row-block path) ----------
__global__ void k_adam_lerp(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            float* __restrict__ tgt, long n, float lr,
                            float b1, float b2, float eps, float tau,
                            const Counters* cnt, int is_actor) {
    long long t = is_actor ? cnt->adam_t_actor : cnt->adam_t_critic;
    float bc1 = 1.f - __powf(b1, (float)t);
    float bc2 = 1.f - __powf(b2, (float)t);
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        float gi = g[i];
        float mi = b1 * m[i] + (1.f - b1) * gi;
        float vi = b2 * v[i] + (1.f - b2) * gi * gi;
        m[i] = mi; v[i] = vi;
        float pn = p[i] - lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
        p[i] = pn;
        tgt[i] += tau * (pn - tgt[i]);
    }
}

// ---- target soft update (K9): one kernel, two slabs ------------------------
__global__ void k_soft_update(float* __restrict__ ta,
                              const float* __restrict__ sa, long na,
                              float* __restrict__ tc,
                              const float* __restrict__ sc, long nc,
                              float tau) {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < na + nc;
         i += (long)gridDim.x * blockDim.x) {
        if (i < na) ta[i] += tau * (sa[i] - ta[i]);
        else { long k = i - na; tc[k] += tau * (sc[k] - tc[k]); }
    }
}

// ---- PER priority write-back (K12 update path) ------------------------------
// Exact level-synchronized tree repair: ONE workgroup; thread i owns probe i;
// per level all touched parents are recomputed from their (already final)
// children.  Duplicate parents write identical values (benign).
__global__ void k_per_update(double* __restrict__ sum_tree,
                             double* __restrict__ min_tree,
                             long tree_cap,
                             const long* __restrict__ idx,
                             const float* __restrict__ pri,
                             int B, float alpha, Counters* cnt) {
    int tid = threadIdx.x;
    // leaves
    float local_max = 0.f;
    for (int i = tid; i < B; i += blockDim.x) {
        float p = pri[i];
        double pa = pow((double)p, (double)alpha);
        long leaf = tree_cap + idx[i];
        sum_tree[leaf] = pa;
        min_tree[leaf] = pa;
        local_max = fmaxf(local_max, p);
    }
    // wg-reduce max priority
    __shared__ float smax[256];
    smax[tid] = local_max;
    __syncthreads();
    for (int s = blockDim.x / 2; s > 0; s >>= 1) {
        if (tid < s) smax[tid] = fmaxf(smax[tid], smax[tid + s]);
        __syncthreads();
    }
    if (tid == 0)
        cnt->max_priority = fmaxf(cnt->max_priority, smax[0]);
    __syncthreads();
    // level-synchronized repair
    long levels = 0;
    for (long c = tree_cap; c > 1; c >>= 1) ++levels;
    for (long lv = 0; lv < levels; ++lv) {
        for (int i = tid; i < B; i += blockDim.x) {
            long node = (tree_cap + idx[i]) >> (lv + 1);
            if (node >= 1) {
                sum_tree[node] = sum_tree[2 * node] + sum_tree[2 * node + 1];
                min_tree[node] = fmin(min_tree[2 * node],
                                      min_tree[2 * node + 1]);
            }
        }
        __syncthreads();
    }
    // last kernel of the step: advance the schedule counters for the NEXT
    // step (counters are initialized to 1 so the first step sees t=1)
    if (tid == 0) {
        cnt->beta_t += 1;
        cnt->adam_t_actor += 1;
        cnt->adam_t_critic += 1;
        cnt->rng_epoch += 1;
    }
}

// ---- PER write-back, wide-batch variant -------------------------------------
// At B >= 512 the single-workgroup level-synced k_per_update serializes
// (measured 305 us at B=4096); instead: one leaves kernel + one kernel per
// tree level (grid-wide), + a tiny counter-tick kernel.  Duplicate parents
// recompute identical values (benign, same as the in-wg version).
__global__ void k_per_leaves(double* __restrict__ sum_tree,
                             double* __restrict__ min_tree, long tree_cap,
                             const long* __restrict__ idx,
                             const float* __restrict__ pri,
                             int B, float alpha, Counters* cnt) {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < B;
         i += (long)gridDim.x * blockDim.x) {
        float p = pri[i];
        double pa = pow((double)p, (double)alpha);
        long leaf = tree_cap + idx[i];
        sum_tree[leaf] = pa;
        min_tree[leaf] = pa;
        // positive floats compare correctly as int bits
        atomicMax((int*)&cnt->max_priority, __float_as_int(p));
    }
}

__global__ void k_per_level(double* __restrict__ sum_tree,
                            double* __restrict__ min_tree, long tree_cap,
                            const long* __restrict__ idx, int B, long lv) {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < B;
         i += (long)gridDim.x * blockDim.x) {
        long node = (tree_cap + idx[i]) >> (lv + 1);
        if (node >= 1) {
            sum_tree[node] = sum_tree[2 * node] + sum_tree[2 * node + 1];
            min_tree[node] = fmin(min_tree[2 * node], min_tree[2 * node + 1]);
        }
    }
}

// Multi-level path repair: writes heights h0+1 .. h0+nlv along each
// sampled path, every node recomputed directly from the already-correct
// height-h0 snapshot (node at height h0+1+j sums its 2^(j+1) CONTIGUOUS
// height-h0 descendants), so no ordering is needed between the levels
// written by one launch.  Replaces nlv small dispatches (~4.8 us kernel
// floor each on this part) with one.
__global__ void k_per_level4(double* __restrict__ sum_tree,
                             double* __restrict__ min_tree, long tree_cap,
                             const long* __restrict__ idx, int B, long h0,
                             int nlv) {
    for (long e = (long)blockIdx.x * blockDim.x + threadIdx.x;
         e < (long)B * nlv; e += (long)gridDim.x * blockDim.x) {
        long i = e / nlv;
        int j = (int)(e % nlv);
        long base = (tree_cap + idx[i]) >> h0;     // height-h0 ancestor
        long node = base >> (j + 1);
        if (node < 1) continue;
        long c0 = node << (j + 1);
        int cnt = 1 << (j + 1);
        // pairwise (binary-tree) combination order, so the written value
        // is BITWISE what the per-level repair would produce — the 4-ary
        // descent depends on parent == sum(children) exactly
        double sv[16], mv[16];
        for (int c = 0; c < cnt; ++c) {
            sv[c] = sum_tree[c0 + c];
            mv[c] = min_tree[c0 + c];
        }
        for (int w = cnt; w > 1; w >>= 1)
            for (int t = 0; t < (w >> 1); ++t) {
                sv[t] = sv[2 * t] + sv[2 * t + 1];
                mv[t] = fmin(mv[2 * t], mv[2 * t + 1]);
            }
        sum_tree[node] = sv[0];
        min_tree[node] = mv[0];
    }
}

__global__ void k_tick_end(Counters* cnt) {
    if (threadIdx.x == 0) {
        cnt->beta_t += 1;
        cnt->adam_t_actor += 1;
        cnt->adam_t_critic += 1;
        cnt->rng_epoch += 1;
    }
}

// ---- replay ingestion (batched add) -----------------------------------------
// T transitions appended at the ring position with priority max_priority^alpha,
// then one level-synced repair pass.  Single workgroup (T can exceed threads).
__global__ void k_replay_add(float* __restrict__ rs, float* __restrict__ ra,
                             float* __restrict__ rr, float* __restrict__ rs2,
                             float* __restrict__ rd,
                             double* __restrict__ sum_tree,
                             double* __restrict__ min_tree,
                             long tree_cap, long capacity,
                             const float* __restrict__ ts,
                             const float* __restrict__ ta,
                             const float* __restrict__ tr,
                             const float* __restrict__ ts2,
                             const float* __restrict__ td,
                             int T, int obs, int act, float alpha,
                             Counters* cnt) {
    int tid = threadIdx.x;
    long pos0 = cnt->pos;
    double pa = pow((double)cnt->max_priority, (double)alpha);
    // copy rows (grid-stride inside the WG over T*max(obs,act) elems)
    for (long e = tid; e < (long)T * obs; e += blockDim.x) {
        long t = e / obs, k = e % obs;
        long slot = (pos0 + t) % capacity;
        rs[slot * obs + k] = ts[t * obs + k];
        rs2[slot * obs + k] = ts2[t * obs + k];
    }
    for (long e = tid; e < (long)T * act; e += blockDim.x) {
        long t = e / act, k = e % act;
        long slot = (pos0 + t) % capacity;
        ra[slot * act + k] = ta[t * act + k];
    }
    for (int t = tid; t < T; t += blockDim.x) {
        long slot = (pos0 + t) % capacity;
        rr[slot] = tr[t];
        rd[slot] = td[t];
        sum_tree[tree_cap + slot] = pa;
        min_tree[tree_cap + slot] = pa;
    }
    __syncthreads();
    long levels = 0;
    for (long c = tree_cap; c > 1; c >>= 1) ++levels;
    for (long lv = 0; lv < levels; ++lv) {
        for (int t = tid; t < T; t += blockDim.x) {
            long slot = (pos0 + t) % capacity;
            long node = (tree_cap + slot) >> (lv + 1);
            if (node >= 1) {
                sum_tree[node] = sum_tree[2 * node] + sum_tree[2 * node + 1];
                min_tree[node] = fmin(min_tree[2 * node],
                                      min_tree[2 * node + 1]);
            }
        }
        __syncthreads();
    }
    if (tid == 0) {
        cnt->pos = (pos0 + T) % capacity;
        cnt->size = cnt->size + T > capacity ? capacity : cnt->size + T;
    }
}

// Bulk variant of the row-copy + leaf-init half of k_replay_add: grid-
// stride over T rows (the one-workgroup form serializes at large T —
// GPU-rollout actor ranks push 100k+ transitions per exchange).  The
// caller repairs the tree with the k_tree_build_level sweep (bandwidth-
// cheap) and bumps the counters host-side.
__global__ void k_replay_copy(float* __restrict__ rs, float* __restrict__ ra,
                              float* __restrict__ rr,
                              float* __restrict__ rs2,
                              float* __restrict__ rd,
                              double* __restrict__ sum_tree,
                              double* __restrict__ min_tree,
                              long tree_cap, long capacity, long pos0,
                              const float* __restrict__ ts,
                              const float* __restrict__ ta,
                              const float* __restrict__ tr,
                              const float* __restrict__ ts2,
                              const float* __restrict__ td,
                              int T, int obs, int act, double pa) {
    for (long e = (long)blockIdx.x * blockDim.x + threadIdx.x;
         e < (long)T * obs; e += (long)gridDim.x * blockDim.x) {
        long t = e / obs, k = e % obs;
        long slot = (pos0 + t) % capacity;
        rs[slot * obs + k] = ts[t * obs + k];
        rs2[slot * obs + k] = ts2[t * obs + k];
    }
    for (long e = (long)blockIdx.x * blockDim.x + threadIdx.x;
         e < (long)T * act; e += (long)gridDim.x * blockDim.x) {
        long t = e / act, k = e % act;
        long slot = (pos0 + t) % capacity;
        ra[slot * act + k] = ta[t * act + k];
    }
    for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < T;
         t += (long)gridDim.x * blockDim.x) {
        long slot = (pos0 + t) % capacity;
        rr[slot] = tr[t];
        rd[slot] = td[t];
        sum_tree[tree_cap + slot] = pa;
        min_tree[tree_cap + slot] = pa;
    }
}

// ---- synthetic replay fill (bench path: no H2D needed) ----------------------
__global__ void k_synth_fill(float* rs, float* ra, float* rr, float* rs2,
                             float* rd, double* sum_tree, double* min_tree,
                             long tree_cap, long capacity, long n,
                             int obs, int act, uint64_t seed, float alpha) {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        Philox4 r = philox4(seed, 0x5EEDull, (uint64_t)i);
        uint32_t st = r.v[0];
        for (int k = 0; k < obs; ++k) {
            Philox4 q = philox4(seed, 1 + (uint64_t)k, (uint64_t)i);
            rs[i * obs + k] = 2.f * u01(q.v[0]) - 1.f;
            rs2[i * obs + k] = 2.f * u01(q.v[1]) - 1.f;
        }
        for (int k = 0; k < act; ++k) {
            Philox4 q = philox4(seed, 1000 + (uint64_t)k, (uint64_t)i);
            ra[i * act + k] = 2.f * u01(q.v[2]) - 1.f;
        }
        rr[i] = -u01(r.v[1]) * 10.f;
        rd[i] = (u01(r.v[2]) < 0.01f) ? 1.f : 0.f;
        sum_tree[tree_cap + i] = 1.0;     // max_priority(1)^alpha
        min_tree[tree_cap + i] = 1.0;
    }
}

__global__ void k_fill_f64(double* p, long n, double v) {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x)
        p[i] = v;
}

// full-tree rebuild after bulk fill: one level at a time, many WGs
__global__ void k_tree_build_level(double* sum_tree, double* min_tree,
                                   long lo, long hi) {
    for (long n = lo + (long)blockIdx.x * blockDim.x + threadIdx.x; n < hi;
         n += (long)gridDim.x * blockDim.x) {
        sum_tree[n] = sum_tree[2 * n] + sum_tree[2 * n + 1];
        min_tree[n] = fmin(min_tree[2 * n], min_tree[2 * n + 1]);
    }
}

// ===========================================================================
// GPU-resident actor rollout: M vectorized Pendulum envs + exploration
// noise (K11) + n-step fold (K15) entirely on device, transitions written
// straight into the on-HBM replay ring.  Re-expresses the reference's
// per-env-step actor loop (/root/reference/main.py:142-152 +
// random_process.py:4-45) and the host twins VectorPendulum / VecNStep
// (d4pg_amd/envs/vector.py), which serve as the parity oracles.
//
// Per tick: [fwd chain obs->act over M rows] then ONE k_roll_tick kernel
// (noise + dynamics + ring write + n-step emit).  Tree leaves are written
// at emit; internal nodes are rebuilt ONCE per episode by the full
// per-level sweep (k_tree_build_level — ~2M f64 nodes, bandwidth-cheap),
// which beats per-tick path repair by orders of magnitude.  The whole
// episode is one hipGraph; per-episode variability (philox epoch, replay
// base position) lives in device memory so graph replays stay fresh.
// ===========================================================================

__device__ inline float d4pg_angle_norm(float x) {
    const float PI = 3.14159265358979323846f;
    float y = fmodf(x + PI, 2.0f * PI);
    if (y < 0.0f) y += 2.0f * PI;
    return y - PI;
}

// standard normal via Box-Muller on two philox lanes
__device__ inline float n01(uint32_t a, uint32_t b) {
    float u1 = u01(a), u2 = u01(b);
    u1 = fmaxf(u1, 1e-12f);
    return sqrtf(-2.0f * logf(u1)) *
           cosf(6.283185307179586f * u2);
}

struct RollArgs {
    int M, O, A, n, horizon;
    float gamma, eps, ou_theta, ou_sigma, ou_mu, ou_dt;
    int noise_kind;                    // 0 = gaussian, 1 = OU
    uint64_t seed;
    float per_alpha;
    // env + noise state
    float *th, *thdot, *obs, *act, *ou_x;
    // n-step rings [n][M][*]
    float *s_ring, *a_ring, *r_ring;
    // per-episode device state: [0] philox epoch, [1] replay base pos
    long long* ep_state;
    // replay
    float *rs, *ra, *rr, *rs2, *rd;
    double *sum_tree, *min_tree;
    long tree_cap, capacity;
    Counters* cnt;
};

// episode prologue: bump the philox epoch, latch the replay write base
__global__ void k_roll_begin(RollArgs a) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        a.ep_state[0] += 1;
        a.ep_state[1] = a.cnt->pos;
    }
}

// reset all M envs (uniform th in [-pi,pi], thdot in [-1,1] — Pendulum-v1
// reset distribution) + OU state to mu, and emit the first observation
__global__ void k_roll_reset(RollArgs a) {
    long long ep = a.ep_state[0];
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < a.M;
         i += gridDim.x * blockDim.x) {
        Philox4 r = philox4(a.seed ^ 0xD011Eull, ((uint64_t)ep << 20) | 1u,
                            (uint64_t)i);
        float th = (2.0f * u01(r.v[0]) - 1.0f) * 3.14159265358979f;
        float td = 2.0f * u01(r.v[1]) - 1.0f;
        a.th[i] = th;
        a.thdot[i] = td;
        a.obs[i * 3 + 0] = cosf(th);
        a.obs[i * 3 + 1] = sinf(th);
        a.obs[i * 3 + 2] = td;
        for (int k = 0; k < a.A; ++k) a.ou_x[i * a.A + k] = a.ou_mu;
    }
}

// one synchronized tick for all M envs: noise -> clip -> dynamics ->
// n-step ring write -> (optional) matured-transition emit into the replay.
// slot/emit_k/done are per-tick constants baked into the episode graph.
__global__ void k_roll_tick(RollArgs a, int tick, int slot, int emit_k,
                            int done) {
    long long ep = a.ep_state[0];
    long long base = a.ep_state[1];
    double pa = pow((double)a.cnt->max_priority, (double)a.per_alpha);
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < a.M;
         i += gridDim.x * blockDim.x) {
        // --- exploration noise (K11): per-env per-tick philox stream ---
        float u_n = a.act[i * a.A];        // policy output, tanh in (-1,1)
        if (a.noise_kind == 0 && a.eps != 0.0f) {
            Philox4 r = philox4(a.seed ^ 0x2F01Eull,
                                ((uint64_t)ep << 20) | (uint64_t)(tick + 2),
                                (uint64_t)i);
            u_n += a.eps * n01(r.v[0], r.v[1]);
        } else if (a.noise_kind == 1) {
            Philox4 r = philox4(a.seed ^ 0x2F01Eull,
                                ((uint64_t)ep << 20) | (uint64_t)(tick + 2),
                                (uint64_t)i);
            float x = a.ou_x[i * a.A];
            x += a.ou_theta * (a.ou_mu - x) * a.ou_dt
                 + a.ou_sigma * sqrtf(a.ou_dt) * n01(r.v[0], r.v[1]);
            a.ou_x[i * a.A] = x;
            u_n += a.eps * x;
        }
        u_n = fminf(fmaxf(u_n, -1.0f), 1.0f);

        // --- n-step ring: store s_t, a_t BEFORE stepping ---
        a.s_ring[(slot * a.M + i) * 3 + 0] = a.obs[i * 3 + 0];
        a.s_ring[(slot * a.M + i) * 3 + 1] = a.obs[i * 3 + 1];
        a.s_ring[(slot * a.M + i) * 3 + 2] = a.obs[i * 3 + 2];
        a.a_ring[slot * a.M + i] = u_n;

        // --- Pendulum dynamics (VectorPendulum.step parity) ---
        const float max_torque = 2.0f, max_speed = 8.0f, dt = 0.05f,
                    g = 10.0f;
        float th = a.th[i], td = a.thdot[i];
        float u = u_n * max_torque;
        float an = d4pg_angle_norm(th);
        float cost = an * an + 0.1f * td * td + 0.001f * u * u;
        float newtd = td + (1.5f * g * sinf(th) + 3.0f * u) * dt;
        newtd = fminf(fmaxf(newtd, -max_speed), max_speed);
        th = th + newtd * dt;
        a.th[i] = th;
        a.thdot[i] = newtd;
        a.r_ring[slot * a.M + i] = -cost;
        float o0 = cosf(th), o1 = sinf(th), o2 = newtd;
        a.obs[i * 3 + 0] = o0;
        a.obs[i * 3 + 1] = o1;
        a.obs[i * 3 + 2] = o2;

        // --- matured n-step emit (VecNStep parity): transition
        // (s_{t-n+1}, a_{t-n+1}, sum gamma^k r, s_{t+1}, done) ---
        if (emit_k >= 0) {
            int start = (slot + 1) % a.n;       // oldest ring entry
            float R = 0.0f, gk = 1.0f;
            for (int k = 0; k < a.n; ++k) {
                R += gk * a.r_ring[((start + k) % a.n) * a.M + i];
                gk *= a.gamma;
            }
            long dst = (base + (long)emit_k * a.M + i) % a.capacity;
            a.rs[dst * 3 + 0] = a.s_ring[(start * a.M + i) * 3 + 0];
            a.rs[dst * 3 + 1] = a.s_ring[(start * a.M + i) * 3 + 1];
            a.rs[dst * 3 + 2] = a.s_ring[(start * a.M + i) * 3 + 2];
            a.ra[dst] = a.a_ring[start * a.M + i];
            a.rr[dst] = R;
            a.rs2[dst * 3 + 0] = o0;
            a.rs2[dst * 3 + 1] = o1;
            a.rs2[dst * 3 + 2] = o2;
            a.rd[dst] = (float)done;
            a.sum_tree[a.tree_cap + dst] = pa;
            a.min_tree[a.tree_cap + dst] = pa;
        }
    }
}

// episode epilogue: advance the replay counters by the emitted block
__global__ void k_roll_end(RollArgs a, long emitted) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        long long base = a.ep_state[1];
        a.cnt->pos = (base + emitted) % a.capacity;
        long long sz = a.cnt->size + emitted;
        a.cnt->size = sz > a.capacity ? a.capacity : sz;
    }
}

// ===========================================================================
// MFMA tiled GEMM kernels (wide-batch learner path, f32-in matrix cores)
// ===========================================================================
// For the large-batch config (BASELINE config 5: B=4096, H=1024) the GEMMs
// are compute-bound and belong on the matrix cores.  gfx950's f32-in MFMA
// (v_mfma_f32_32x32x2_f32) runs at the full fp32 rate (157 TF peak) with
// EXACT f32 numerics — bitwise a k-ordered fmaf chain, i.e. the same
// summation order as the per-thread loops of the small-batch kernels.
// Tile: 128x128 per workgroup (4 waves, each 64x64 = 2x2 of 32x32 MFMA
// tiles), K staged through LDS in 32-deep slices.
// M-tile 64 (not 128): the H=1024 layers' grids then cover the chip with
// 2 workgroups per CU, so one workgroup's LDS staging hides behind the
// other's MFMAs (at 128 the 256-wg grid left 1 wg/CU and the per-chunk
// ds_write+barrier was exposed — ~50% MfmaUtil).
#define MT_M 64
#define MT_N 128
#define MT_K 32

typedef float f32x16 __attribute__((ext_vector_type(16)));

// Software-pipelined MFMA tile core shared by fwd/dX/dW: double-buffered
// LDS (global loads for chunk ch+1 are issued before chunk ch's MFMAs and
// written to the other buffer after them — one barrier per chunk), with
// register-batched 16-load staging so every global load is in flight at
// once.  `ldA`/`ldB` load chunk k0's A/B fragments into 16 registers.
// If `bias_acc` is non-null, B-fragment values for waves with wm0==0 are
// summed into bias_acc[0..1] during the MFMA loop (used by dW for db).
template <typename FA, typename FB>
__device__ inline void mfma_pipeline(FA ldA, FB ldB, int nch,
                                     float* As0, float* As1,
                                     float* Bs0, float* Bs1,
                                     f32x16& a00, f32x16& a01,
                                     float* bias_acc) {
    int tid = threadIdx.x, wid = tid >> 6, lane = tid & 63;
    int wm0 = (wid >> 1) * 32, wn0 = (wid & 1) * 64;
    int r = lane & 31, kk2 = lane >> 5;
    float ta[8], tb[16];
    ldA(0, ta);
    ldB(0, tb);
    auto wr = [&](float* As, float* Bs) {
#pragma unroll
        for (int u = 0; u < 8; ++u) {
            int e = u * 256 + tid;
            As[(e & 31) * (MT_M + 4) + (e >> 5)] = ta[u];
        }
#pragma unroll
        for (int u = 0; u < 16; ++u) {
            int e = u * 256 + tid;
            Bs[(e >> 7) * (MT_N + 4) + (e & 127)] = tb[u];
        }
    };
    wr(As0, Bs0);
    __syncthreads();
    for (int ch = 0; ch < nch; ++ch) {
        float* As = (ch & 1) ? As1 : As0;
        float* Bs = (ch & 1) ? Bs1 : Bs0;
        bool more = ch + 1 < nch;
        if (more) {
            ldA((ch + 1) * MT_K, ta);
            ldB((ch + 1) * MT_K, tb);
        }
#pragma unroll
        for (int ks = 0; ks < MT_K; ks += 2) {
            float a0 = As[(ks + kk2) * (MT_M + 4) + wm0 + r];
            float b0 = Bs[(ks + kk2) * (MT_N + 4) + wn0 + r];
            float b1 = Bs[(ks + kk2) * (MT_N + 4) + wn0 + 32 + r];
            a00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, a00, 0, 0, 0);
            a01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, a01, 0, 0, 0);
            if (bias_acc && wm0 == 0) {
                bias_acc[0] += b0;
                bias_acc[1] += b1;
            }
        }
        if (more) wr((ch & 1) ? As0 : As1, (ch & 1) ? Bs0 : Bs1);
        __syncthreads();
    }
}

// Variant of mfma_pipeline with caller-supplied register->LDS writer and
// VECTORIZED staging: the staging instruction stream (scalar loads +
// per-element bounds checks + scattered ds_writes) of the original
// closures was ~350 VALU-pipe instructions per chunk — comparable to the
// chunk's 32 MFMAs' ~1400 MAI cycles — capping the H-GEMMs at ~44% of
// the measured 156 TF/s f32 MFMA peak (scripts/mfma_peak,
// profiles/gemm_lab).  Interior tiles load float4 and write b128 rows,
// cutting staging to ~100 instructions; edge tiles keep the guarded
// scalar path.  LDS contents are bit-identical to mfma_pipeline's.
template <typename FA, typename FB, typename FW, typename FWS>
__device__ inline void mfma_pipeline_w(FA ldA, FB ldB, FW wr, FWS wrs,
                                       int nch, float* As0, float* As1,
                                       float* Bs0, float* Bs1, f32x16& a00,
                                       f32x16& a01) {
    int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
    int wm0 = (wid >> 1) * 32, wn0 = (wid & 1) * 64;
    int r = lane & 31, kk2 = lane >> 5;
    ldA(0);
    ldB(0);
    wr(As0, Bs0);
    __syncthreads();
    for (int ch = 0; ch < nch; ++ch) {
        float* As = (ch & 1) ? As1 : As0;
        float* Bs = (ch & 1) ? Bs1 : Bs0;
        float* Asn = (ch & 1) ? As0 : As1;
        float* Bsn = (ch & 1) ? Bs0 : Bs1;
        bool more = ch + 1 < nch;
        if (more) {
            ldA((ch + 1) * MT_K);
            ldB((ch + 1) * MT_K);
        }
        // next chunk's LDS writes ride the MFMA shadows as slices 6..15
        // (slice 6 trails the loads by ~500 MFMA-pipe cycles, past L2
        // latency) instead of bunching at the barrier, where the two
        // co-resident workgroups' bursts convoy and expose the staging
#pragma unroll
        for (int ks = 0; ks < MT_K; ks += 2) {
            float a0 = As[(ks + kk2) * (MT_M + 4) + wm0 + r];
            float b0 = Bs[(ks + kk2) * (MT_N + 4) + wn0 + r];
            float b1 = Bs[(ks + kk2) * (MT_N + 4) + wn0 + 32 + r];
            a00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, a00, 0, 0, 0);
            a01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, a01, 0, 0, 0);
            if (more) wrs(ks >> 1, Asn, Bsn);
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// Big-GEMM variant: 128x256 workgroup tiles + split-K.
// Tile-traffic accounting (NOTES.md): HBM bytes scale with
// (1/MT_M + 1/MT_N), so 128x256 halves the weight re-streaming of 64x128
// (402 -> 201 MB per 4096x1024x1024 GEMM) while split-K keeps the grid at
// >= 256 workgroups for full MFMA issue. Partial sums land in a zeroed
// scratch via fp32 atomics; a grid-stride epilogue applies bias+activation
// (or the backward mask).
#define M2_M 128
#define M2_N 256

template <typename FA, typename FB>
__device__ inline void mfma_pipeline2(FA ldA, FB ldB, int nch,
                                      float* As0, float* As1,
                                      float* Bs0, float* Bs1,
                                      f32x16 acc[2][4]) {
    int tid = threadIdx.x, wid = tid >> 6, lane = tid & 63;
    int wm0 = (wid >> 1) * 64, wn0 = (wid & 1) * 128;
    int r = lane & 31, kk2 = lane >> 5;
    float ta[16], tb[32];
    ldA(0, ta);
    ldB(0, tb);
    auto wr = [&](float* As, float* Bs) {
#pragma unroll
        for (int u = 0; u < 16; ++u) {
            int e = u * 256 + tid;
            As[(e & 31) * (M2_M + 4) + (e >> 5)] = ta[u];
        }
#pragma unroll
        for (int u = 0; u < 32; ++u) {
            int e = u * 256 + tid;
            Bs[(e >> 8) * (M2_N + 4) + (e & 255)] = tb[u];
        }
    };
    wr(As0, Bs0);
    __syncthreads();
    for (int ch = 0; ch < nch; ++ch) {
        float* As = (ch & 1) ? As1 : As0;
        float* Bs = (ch & 1) ? Bs1 : Bs0;
        bool more = ch + 1 < nch;
        if (more) {
            ldA((ch + 1) * MT_K, ta);
            ldB((ch + 1) * MT_K, tb);
        }
#pragma unroll
        for (int ks = 0; ks < MT_K; ks += 2) {
            float a0 = As[(ks + kk2) * (M2_M + 4) + wm0 + r];
            float a1 = As[(ks + kk2) * (M2_M + 4) + wm0 + 32 + r];
            float b[4];
#pragma unroll
            for (int j = 0; j < 4; ++j)
                b[j] = Bs[(ks + kk2) * (M2_N + 4) + wn0 + 32 * j + r];
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                acc[0][j] = __builtin_amdgcn_mfma_f32_32x32x2f32(
                    a0, b[j], acc[0][j], 0, 0, 0);
                acc[1][j] = __builtin_amdgcn_mfma_f32_32x32x2f32(
                    a1, b[j], acc[1][j], 0, 0, 0);
            }
        }
        if (more) wr((ch & 1) ? As0 : As1, (ch & 1) ? Bs0 : Bs1);
        __syncthreads();
    }
}

#define MFMA2_LDS_DECL \
    __shared__ float As2[2][MT_K * (M2_M + 4)]; \
    __shared__ float Bs2[2][MT_K * (M2_N + 4)]

// partial C[B,out] += X-chunk @ W-chunk (split-K over `ksplit` segments,
// fp32-atomic accumulation into pre-zeroed `part`)
__global__ void __launch_bounds__(256, 1)
k_mfma_fwd2(const float* __restrict__ x1, const float* __restrict__ x2,
            const float* __restrict__ wt, float* __restrict__ part,
            int B, int in1, int in2, int out, int ksplit) {
    MFMA2_LDS_DECL;
    int in_total = in1 + in2;
    int ntm = (B + M2_M - 1) / M2_M;
    int ntn = (out + M2_N - 1) / M2_N;
    int seg = blockIdx.x / (ntm * ntn);
    int tile = blockIdx.x % (ntm * ntn);
    int m0 = (tile / ntn) * M2_M, n0 = (tile % ntn) * M2_N;
    int nch_total = (in_total + MT_K - 1) / MT_K;
    int nch_seg = (nch_total + ksplit - 1) / ksplit;
    int ch0 = seg * nch_seg;
    int nch = min(nch_seg, nch_total - ch0);
    if (nch <= 0) return;
    int k_base = ch0 * MT_K;
    int tid = threadIdx.x, lane = tid & 63;
    f32x16 acc[2][4] = {};
    auto ldA = [&](int k0, float* t) {
#pragma unroll
        for (int u = 0; u < 16; ++u) {
            int e = u * 256 + tid;
            int gm = m0 + (e >> 5), gk = k_base + k0 + (e & 31);
            float v = 0.f;
            if (gm < B && gk < in_total)
                v = (gk < in1) ? x1[(long)gm * in1 + gk]
                               : x2[(long)gm * in2 + (gk - in1)];
            t[u] = v;
        }
    };
    auto ldB = [&](int k0, float* t) {
#pragma unroll
        for (int u = 0; u < 32; ++u) {
            int e = u * 256 + tid;
            int gk = k_base + k0 + (e >> 8), gn = n0 + (e & 255);
            t[u] = (gk < in_total && gn < out)
                ? wt[(long)gk * out + gn] : 0.f;
        }
    };
    mfma_pipeline2(ldA, ldB, nch, As2[0], As2[1], Bs2[0], Bs2[1], acc);
    int wid = tid >> 6;
    int wm0 = (wid >> 1) * 64, wn0 = (wid & 1) * 128;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int reg = 0; reg < 16; ++reg) {
                int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
                int col = lane & 31;
                int gm = m0 + wm0 + i * 32 + row;
                int gn = n0 + wn0 + j * 32 + col;
                if (gm < B && gn < out)
                    atomicAdd(&part[(long)gm * out + gn], acc[i][j][reg]);
            }
        }
    }
}

// epilogue: y = act(part + bias) (fwd) or dx = part * act_mask (dX)
__global__ void k_mfma_epilogue(const float* __restrict__ part,
                                const float* __restrict__ bias,
                                const float* __restrict__ hprev,
                                float* __restrict__ y, long n, int out,
                                int act_kind, int prev_act) {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        float v = part[i];
        if (bias) {
            v += bias[i % out];
            if (act_kind == ACT_RELU) v = fmaxf(v, 0.f);
            else if (act_kind == ACT_TANH) v = tanhf(v);
        } else if (hprev) {
            v *= act_mask(prev_act, hprev[i]);
        }
        y[i] = v;
    }
}

#define MFMA_LDS_DECL \
    __shared__ float As2[2][MT_K * (MT_M + 4)]; \
    __shared__ float Bs2[2][MT_K * (MT_N + 4)]

// C[B,out] = act(X[B,in1 (++ concat in2)] @ Wt[in,out] + bias).
// ksplit > 1: split-K over `parts` (disjoint per-segment partials summed
// by k_head_finish) — used for the narrow heads (out <= 64), whose
// M-tile-only grids (64 wgs) otherwise leave 3/4 of the chip idle.
__global__ void __launch_bounds__(256, 2)
k_mfma_fwd(const float* __restrict__ x1, const float* __restrict__ x2,
           const float* __restrict__ wt, const float* __restrict__ bias,
           float* __restrict__ y, int B, int in1, int in2, int out,
           int act_kind, int ksplit, float* __restrict__ parts) {
    MFMA_LDS_DECL;
    int in_total = in1 + in2;
    int ntn = (out + MT_N - 1) / MT_N;
    int ntm = (B + MT_M - 1) / MT_M;
    int seg = blockIdx.x / (ntm * ntn);
    int tile = blockIdx.x % (ntm * ntn);
    int m0 = (tile / ntn) * MT_M, n0 = (tile % ntn) * MT_N;
    int nch_total = (in_total + MT_K - 1) / MT_K;
    int nch_seg = (nch_total + ksplit - 1) / ksplit;
    int ch0 = seg * nch_seg;
    int nch = min(nch_seg, nch_total - ch0);
    int kbase = ch0 * MT_K;
    int tid = threadIdx.x, lane = tid & 63;
    f32x16 acc00 = {}, acc01 = {};
    // vectorized staging geometry (see mfma_pipeline_w):
    //   A: thread owns row m0+tid/4, k-segment (tid%4)*8 -> 2x float4
    //      loads, scalar transposed LDS writes
    //   B: thread owns k-row tid/8, n-segment (tid%8)*16 -> 4x float4
    //      loads AND 4x b128 LDS writes (row-major, no transpose)
    const int am = tid >> 2, ak = (tid & 3) * 8;
    const int bk = tid >> 3, bn = (tid & 7) * 16;
    const bool vecA = (in1 % 4 == 0) && (m0 + MT_M <= B);
    const bool vecB = (out % 4 == 0) && (n0 + MT_N <= out);
    float ta[8];
    float4 tb[4];
    auto ldA = [&](int k0r) {
        int k0 = kbase + k0r;
        if (vecA && k0 + MT_K <= in1) {
            const float* src = x1 + (long)(m0 + am) * in1 + k0 + ak;
            float4 v0 = *reinterpret_cast<const float4*>(src);
            float4 v1 = *reinterpret_cast<const float4*>(src + 4);
            ta[0] = v0.x; ta[1] = v0.y; ta[2] = v0.z; ta[3] = v0.w;
            ta[4] = v1.x; ta[5] = v1.y; ta[6] = v1.z; ta[7] = v1.w;
        } else {
            int gm = m0 + am;
#pragma unroll
            for (int u = 0; u < 8; ++u) {
                int gk = k0 + ak + u;
                float v = 0.f;
                if (gm < B && gk < in_total)
                    v = (gk < in1) ? x1[(long)gm * in1 + gk]
                                   : x2[(long)gm * in2 + (gk - in1)];
                ta[u] = v;
            }
        }
    };
    auto ldB = [&](int k0r) {
        int k0 = kbase + k0r;
        int gk = k0 + bk;
        if (vecB && k0 + MT_K <= in_total) {
            const float4* src = reinterpret_cast<const float4*>(
                wt + (long)gk * out + n0 + bn);
#pragma unroll
            for (int u = 0; u < 4; ++u) tb[u] = src[u];
        } else {
#pragma unroll
            for (int u = 0; u < 16; ++u) {
                int gn = n0 + bn + u;
                float v = (gk < in_total && gn < out)
                    ? wt[(long)gk * out + gn] : 0.f;
                reinterpret_cast<float*>(tb)[u] = v;
            }
        }
    };
    auto wr = [&](float* As, float* Bs) {
#pragma unroll
        for (int u = 0; u < 8; ++u)
            As[(ak + u) * (MT_M + 4) + am] = ta[u];
        float4* dst = reinterpret_cast<float4*>(&Bs[bk * (MT_N + 4) + bn]);
#pragma unroll
        for (int u = 0; u < 4; ++u) dst[u] = tb[u];
    };
    // sliced form for the in-loop pipeline: A scalar writes at slices
    // 6..13, B b128 writes at 12..15 (two streams overlap harmlessly)
    auto wrs = [&](int sl, float* As, float* Bs) {
        if (sl >= 6 && sl < 14)
            As[(ak + sl - 6) * (MT_M + 4) + am] = ta[sl - 6];
        if (sl >= 12)
            reinterpret_cast<float4*>(
                &Bs[bk * (MT_N + 4) + bn])[sl - 12] = tb[sl - 12];
    };
    mfma_pipeline_w(ldA, ldB, wr, wrs, nch, As2[0], As2[1], Bs2[0], Bs2[1],
                    acc00, acc01);
    int wid = tid >> 6;
    int wm0 = (wid >> 1) * 32, wn0 = (wid & 1) * 64;
    const f32x16* accs[2] = {&acc00, &acc01};
#pragma unroll
    for (int t = 0; t < 2; ++t) {
        int i = 0, j = t;
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
            int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
            int col = lane & 31;
            int gm = m0 + wm0 + i * 32 + row;
            int gn = n0 + wn0 + j * 32 + col;
            if (gm < B && gn < out) {
                if (ksplit > 1) {
                    parts[((long)seg * B + gm) * out + gn] =
                        (*accs[t])[reg];
                } else {
                    float v = (*accs[t])[reg] + bias[gn];
                    if (act_kind == ACT_RELU) v = fmaxf(v, 0.f);
                    else if (act_kind == ACT_TANH) v = tanhf(v);
                    y[(long)gm * out + gn] = v;
                }
            }
        }
    }
}

// Narrow dX (span <= 8, the concat action slice): wave per batch row,
// lane-parallel over `out` with shfl reduce.  The MFMA path for this
// shape (one N-tile -> 64-wg grid) left 3/4 of the chip idle and its
// staging fully exposed (~76 us for 50 MFLOP); this VALU form runs the
// whole thing in ~10 us.
__global__ void k_dx_narrow(const float* __restrict__ dz,
                            const float* __restrict__ wt,
                            const float* __restrict__ ymask,
                            float* __restrict__ dx, int B, int in_lo,
                            int in_hi, int out, int prev_act) {
    int lane = threadIdx.x & 63, wrow = threadIdx.x / 64;
    int wpb = blockDim.x / 64;
    int span = in_hi - in_lo;
    for (int row = blockIdx.x * wpb + wrow; row < B;
         row += gridDim.x * wpb) {
        for (int j = 0; j < span; ++j) {
            const float* wrowp = wt + (long)(in_lo + j) * out;
            float acc = 0.f;
            for (int o = lane; o < out; o += 64)
                acc += dz[(long)row * out + o] * wrowp[o];
            for (int s = 32; s > 0; s >>= 1)
                acc += __shfl_xor(acc, s, 64);
            if (lane == 0) {
                float m_ = ymask
                    ? act_mask(prev_act, ymask[(long)row * span + j]) : 1.f;
                dx[(long)row * span + j] = acc * m_;
            }
        }
    }
}

// Head finish: y[row] = act(sum_seg parts[seg][row] + bias), wave == row
// (out <= 64); softmax reduces across the wave like k_fwd3's epilogue.
__global__ void k_head_finish(const float* __restrict__ parts,
                              const float* __restrict__ bias,
                              float* __restrict__ y, int B, int out,
                              int ksplit, int act_kind) {
    int lane = threadIdx.x & 63, wrow = threadIdx.x / 64;
    int wpb = blockDim.x / 64;
    for (int row = blockIdx.x * wpb + wrow; row < B;
         row += gridDim.x * wpb) {
        float v = 0.f;
        if (lane < out) {
            for (int g = 0; g < ksplit; ++g)
                v += parts[((long)g * B + row) * out + lane];
            v += bias[lane];
        }
        if (act_kind == ACT_SOFTMAX) {
            float mx = (lane < out) ? v : -INFINITY;
            for (int s = 32; s > 0; s >>= 1)
                mx = fmaxf(mx, __shfl_xor(mx, s, 64));
            float e = (lane < out) ? __expf(v - mx) : 0.f;
            float sum = e;
            for (int s = 32; s > 0; s >>= 1)
                sum += __shfl_xor(sum, s, 64);
            if (lane < out) y[(long)row * out + lane] = e / sum;
        } else if (lane < out) {
            if (act_kind == ACT_RELU) v = fmaxf(v, 0.f);
            else if (act_kind == ACT_TANH) v = tanhf(v);
            y[(long)row * out + lane] = v;
        }
    }
}

// dX[B, in_lo:in_hi] = (dz[B,out] @ Wt^T) * act'(hprev); hprev/dx are
// [B][span] (span = in_hi - in_lo), hprev null => no mask.
__global__ void __launch_bounds__(256, 2)
k_mfma_dx(const float* __restrict__ dz, const float* __restrict__ wt,
          const float* __restrict__ hprev, float* __restrict__ dx,
          int B, int in_lo, int in_hi, int out, int prev_act) {
    MFMA_LDS_DECL;
    int span = in_hi - in_lo;
    int ntn = (span + MT_N - 1) / MT_N;
    int m0 = (blockIdx.x / ntn) * MT_M;
    int n0 = (blockIdx.x % ntn) * MT_N;             // relative i tile
    int tid = threadIdx.x, lane = tid & 63;
    f32x16 acc00 = {}, acc01 = {};
    // Vectorized staging (see k_mfma_fwd):
    //   A[k(=o)][m(=b)] = dz[b][o]: thread owns dz-row m0+tid/4,
    //     o-segment (tid%4)*8 -> 2x float4 loads, scalar transposed
    //     LDS writes
    //   B[k(=o)][n(=i)] = wt[i][o]: thread owns wt-row in_lo+n0+tid/2,
    //     o-segment (tid%2)*16 -> 4x float4 loads, scalar transposed
    //     LDS writes
    const int am = tid >> 2, ak = (tid & 3) * 8;
    const int bnn = tid >> 1, bkk = (tid & 1) * 16;
    const bool vecA = (out % 4 == 0) && (m0 + MT_M <= B);
    const bool vecB = (out % 4 == 0) && (n0 + MT_N <= span);
    float ta[8], tb[16];
    auto ldA = [&](int k0) {
        if (vecA && k0 + MT_K <= out) {
            const float* src = dz + (long)(m0 + am) * out + k0 + ak;
            float4 v0 = *reinterpret_cast<const float4*>(src);
            float4 v1 = *reinterpret_cast<const float4*>(src + 4);
            ta[0] = v0.x; ta[1] = v0.y; ta[2] = v0.z; ta[3] = v0.w;
            ta[4] = v1.x; ta[5] = v1.y; ta[6] = v1.z; ta[7] = v1.w;
        } else {
            int gm = m0 + am;
#pragma unroll
            for (int u = 0; u < 8; ++u) {
                int gk = k0 + ak + u;
                ta[u] = (gm < B && gk < out)
                    ? dz[(long)gm * out + gk] : 0.f;
            }
        }
    };
    auto ldB = [&](int k0) {
        int gi = in_lo + n0 + bnn;
        if (vecB && k0 + MT_K <= out) {
            const float4* src = reinterpret_cast<const float4*>(
                wt + (long)gi * out + k0 + bkk);
#pragma unroll
            for (int u = 0; u < 4; ++u)
                reinterpret_cast<float4*>(tb)[u] = src[u];
        } else {
#pragma unroll
            for (int u = 0; u < 16; ++u) {
                int gk = k0 + bkk + u;
                tb[u] = (gi < in_hi && gk < out)
                    ? wt[(long)gi * out + gk] : 0.f;
            }
        }
    };
    {
        ldA(0);
        ldB(0);
        auto wr = [&](float* As, float* Bs) {
#pragma unroll
            for (int u = 0; u < 8; ++u)
                As[(ak + u) * (MT_M + 4) + am] = ta[u];
#pragma unroll
            for (int u = 0; u < 16; ++u)
                Bs[(bkk + u) * (MT_N + 4) + bnn] = tb[u];
        };
        wr(As2[0], Bs2[0]);
        __syncthreads();
        int wid = tid >> 6;
        int wm0 = (wid >> 1) * 32, wn0 = (wid & 1) * 64;
        int r = lane & 31, kk2 = lane >> 5;
        int nch = (out + MT_K - 1) / MT_K;
        for (int ch = 0; ch < nch; ++ch) {
            float* As = As2[ch & 1];
            float* Bs = Bs2[ch & 1];
            float* Asn = As2[(ch + 1) & 1];
            float* Bsn = Bs2[(ch + 1) & 1];
            bool more = ch + 1 < nch;
            if (more) {
                ldA((ch + 1) * MT_K);
                ldB((ch + 1) * MT_K);
            }
#pragma unroll
            for (int ks = 0; ks < MT_K; ks += 2) {
                float a0 = As[(ks + kk2) * (MT_M + 4) + wm0 + r];
                float b0 = Bs[(ks + kk2) * (MT_N + 4) + wn0 + r];
                float b1 = Bs[(ks + kk2) * (MT_N + 4) + wn0 + 32 + r];
                acc00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc00, 0, 0, 0);
                acc01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc01, 0, 0, 0);
                // next chunk's LDS writes ride the MFMA shadows: one A
                // element + two B elements per slice over slices 6..13
                if (more) {
                    int sl = ks >> 1;
                    if (sl >= 6 && sl < 14) {
                        int u = sl - 6;
                        Asn[(ak + u) * (MT_M + 4) + am] = ta[u];
                        Bsn[(bkk + 2 * u) * (MT_N + 4) + bnn] = tb[2 * u];
                        Bsn[(bkk + 2 * u + 1) * (MT_N + 4) + bnn] =
                            tb[2 * u + 1];
                    }
                }
            }
            __syncthreads();
        }
    }
    int wid = tid >> 6;
    int wm0 = (wid >> 1) * 32, wn0 = (wid & 1) * 64;
    const f32x16* accs[2] = {&acc00, &acc01};
#pragma unroll
    for (int t = 0; t < 2; ++t) {
        int i = 0, j = t;
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
            int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
            int col = lane & 31;
            int gm = m0 + wm0 + i * 32 + row;
            int gn = n0 + wn0 + j * 32 + col;          // relative i
            if (gm < B && gn < span) {
                long rel = (long)gm * span + gn;
                float m_ = hprev ? act_mask(prev_act, hprev[rel]) : 1.f;
                dx[rel] = (*accs[t])[reg] * m_;
            }
        }
    }
}

// dWt[in,out] = X^T[in,B] @ dz[B,out]  (K dimension = batch).  Split-K
// over `ksplit` segments (grid = nti*nto*ksplit) with fp32 atomicAdd
// epilogue when ksplit > 1 (dwt/db must be pre-zeroed by the caller in
// that case); bias db[o] = sum_b dz[b][o] accumulates for free out of the
// B-fragment during the MFMA loop (wm0==0 waves of the m0==0 tiles).
// Split-K partial reduce: dwt[i,o] = sum_seg parts[seg][i,o] (+ db tail).
// Replaces the fp32-atomicAdd epilogue of the split-K dW: 2.1M atomics per
// H-layer dW cost ~80 us of L2 read-modify-write serialization, while
// disjoint partials + this bandwidth-bound sweep cost ~3 us
// (profiles/gemm_lab evidence).
__global__ void k_dw_reduce(const float* __restrict__ parts,
                            float* __restrict__ dwt, float* __restrict__ db,
                            long nw, int out, int ksplit) {
    long total = nw + (db ? out : 0);
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (long)gridDim.x * blockDim.x) {
        float s = 0.f;
        if (i < nw) {
            for (int g = 0; g < ksplit; ++g)
                s += parts[(long)g * nw + i];
            dwt[i] = s;
        } else {
            long j = i - nw;
            for (int g = 0; g < ksplit; ++g)
                s += parts[(long)ksplit * nw + (long)g * out + j];
            db[j] = s;
        }
    }
}

__global__ void __launch_bounds__(256, 2)
k_mfma_dw(const float* __restrict__ dz, const float* __restrict__ x1,
          const float* __restrict__ x2, float* __restrict__ dwt,
          float* __restrict__ db, int B, int in1, int in2, int out,
          int ksplit, float* __restrict__ parts) {
    MFMA_LDS_DECL;
    int in_total = in1 + in2;
    int ntn = (out + MT_N - 1) / MT_N;
    int nti = (in_total + MT_M - 1) / MT_M;
    int seg = blockIdx.x / (nti * ntn);
    int tile = blockIdx.x % (nti * ntn);
    int m0 = (tile / ntn) * MT_M;                      // i tile
    int n0 = (tile % ntn) * MT_N;                      // o tile
    int nch_total = (B + MT_K - 1) / MT_K;
    int nch_seg = (nch_total + ksplit - 1) / ksplit;
    int ch0 = seg * nch_seg;
    int nch = min(nch_seg, nch_total - ch0);
    if (nch <= 0) return;
    long b_base = (long)ch0 * MT_K;
    int tid = threadIdx.x, lane = tid & 63;
    f32x16 acc00 = {}, acc01 = {};
    // Vectorized staging (same rationale as k_mfma_fwd):
    //   A[k(=b)][m(=i)]: thread owns batch-row b_base+k0+tid/8,
    //     i-segment (tid%8)*8 -> 2x float4 loads AND b128 LDS writes
    //     (no transpose: global and LDS are both i-fastest)
    //   B[k(=b)][n(=o)]: thread owns batch-row tid/8, o-segment
    //     (tid%8)*16 -> 4x float4 loads and b128 writes
    const int akb = tid >> 3, ai = (tid & 7) * 8;
    const int bo = (tid & 7) * 16;
    const bool vecA = (in1 % 4 == 0) && (m0 + MT_M <= in1);
    const bool vecB = (out % 4 == 0) && (n0 + MT_N <= out);
    float4 ta[2], tb[4];
    auto ldA = [&](int k0) {
        long gb = b_base + k0 + akb;
        if (vecA && b_base + k0 + MT_K <= B) {
            const float* src = x1 + gb * in1 + m0 + ai;
            ta[0] = *reinterpret_cast<const float4*>(src);
            ta[1] = *reinterpret_cast<const float4*>(src + 4);
        } else {
#pragma unroll
            for (int u = 0; u < 8; ++u) {
                int gi = m0 + ai + u;
                float v = 0.f;
                if (gb < B && gi < in_total)
                    v = (gi < in1) ? x1[gb * in1 + gi]
                                   : x2[gb * in2 + (gi - in1)];
                reinterpret_cast<float*>(ta)[u] = v;
            }
        }
    };
    auto ldB = [&](int k0) {
        long gb = b_base + k0 + akb;
        if (vecB && b_base + k0 + MT_K <= B) {
            const float4* src =
                reinterpret_cast<const float4*>(dz + gb * out + n0 + bo);
#pragma unroll
            for (int u = 0; u < 4; ++u) tb[u] = src[u];
        } else {
#pragma unroll
            for (int u = 0; u < 16; ++u) {
                int go = n0 + bo + u;
                reinterpret_cast<float*>(tb)[u] =
                    (gb < B && go < out) ? dz[gb * out + go] : 0.f;
            }
        }
    };
    float bias0 = 0.f, bias1 = 0.f;
    {
        int wid = tid >> 6;
        int wm0 = (wid >> 1) * 32, wn0 = (wid & 1) * 64;
        int r = lane & 31, kk2 = lane >> 5;
        ldA(0);
        ldB(0);
        auto wr = [&](float* As, float* Bs) {
            float4* da =
                reinterpret_cast<float4*>(&As[akb * (MT_M + 4) + ai]);
            da[0] = ta[0];
            da[1] = ta[1];
            float4* dbp =
                reinterpret_cast<float4*>(&Bs[akb * (MT_N + 4) + bo]);
#pragma unroll
            for (int u = 0; u < 4; ++u) dbp[u] = tb[u];
        };
        wr(As2[0], Bs2[0]);
        __syncthreads();
        for (int ch = 0; ch < nch; ++ch) {
            float* As = As2[ch & 1];
            float* Bs = Bs2[ch & 1];
            float* Asn = As2[(ch + 1) & 1];
            float* Bsn = Bs2[(ch + 1) & 1];
            bool more = ch + 1 < nch;
            if (more) {
                ldA((ch + 1) * MT_K);
                ldB((ch + 1) * MT_K);
            }
#pragma unroll
            for (int ks = 0; ks < MT_K; ks += 2) {
                float a0 = As[(ks + kk2) * (MT_M + 4) + wm0 + r];
                float b0 = Bs[(ks + kk2) * (MT_N + 4) + wn0 + r];
                float b1 = Bs[(ks + kk2) * (MT_N + 4) + wn0 + 32 + r];
                acc00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc00, 0, 0, 0);
                acc01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc01, 0, 0, 0);
                if (wm0 == 0) { bias0 += b0; bias1 += b1; }
                // next chunk's LDS writes ride the MFMA shadows
                if (more) {
                    int sl = ks >> 1;
                    if (sl == 6)
                        reinterpret_cast<float4*>(
                            &Asn[akb * (MT_M + 4) + ai])[0] = ta[0];
                    else if (sl == 7)
                        reinterpret_cast<float4*>(
                            &Asn[akb * (MT_M + 4) + ai])[1] = ta[1];
                    else if (sl >= 8 && sl < 12)
                        reinterpret_cast<float4*>(
                            &Bsn[akb * (MT_N + 4) + bo])[sl - 8] = tb[sl - 8];
                }
            }
            __syncthreads();
        }
    }
    int wid = tid >> 6;
    int wm0 = (wid >> 1) * 32, wn0 = (wid & 1) * 64;
    const f32x16* accs[2] = {&acc00, &acc01};
#pragma unroll
    for (int t = 0; t < 2; ++t) {
        int i = 0, j = t;
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
            int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
            int col = lane & 31;
            int gi = m0 + wm0 + i * 32 + row;
            int go = n0 + wn0 + j * 32 + col;
            if (gi < in_total && go < out) {
                if (ksplit > 1)
                    // disjoint per-segment partials (k_dw_reduce sums)
                    parts[((long)seg * in_total + gi) * out + go] =
                        (*accs[t])[reg];
                else
                    dwt[(long)gi * out + go] = (*accs[t])[reg];
            }
        }
    }
    if (db && wm0 == 0 && m0 == 0) {
        // combine the two k-parity halves (lane l and l^32 share a column)
        bias0 += __shfl_xor(bias0, 32, 64);
        bias1 += __shfl_xor(bias1, 32, 64);
        int r = lane & 31;
        if ((lane >> 5) == 0) {
            int go0 = n0 + wn0 + r, go1 = n0 + wn0 + 32 + r;
            if (ksplit > 1) {
                long off = (long)ksplit * in_total * out + (long)seg * out;
                if (go0 < out) parts[off + go0] = bias0;
                if (go1 < out) parts[off + go1] = bias1;
            } else {
                if (go0 < out) db[go0] = bias0;
                if (go1 < out) db[go1] = bias1;
            }
        }
    }
}

// ===========================================================================
// Persistent whole-step megakernel (flagship small-batch learner path)
// ===========================================================================
// Rationale (measured, profiles/step_kernel_stats round 1): the row-block
// path spends 451 of 497 us/step in two kernels where each of B=64 waves
// re-streams EVERY layer weight from L2 with only 16 workgroups in flight —
// 64x redundant weight traffic and no latency hiding.  This kernel instead
// runs the WHOLE train step (PER sample -> forwards -> C51 projection ->
// CE/policy grads -> dX/dW backward -> Adam+soft-update -> PER write-back)
// as ONE persistent kernel: PNWG=64 workgroups x 256 threads (guaranteed
// co-resident on 256 CUs, so a software grid barrier is deadlock-free),
// ~28 layer-parallel phases separated by grid barriers.  Each phase tiles
// its GEMM over the grid (16-row x 64-col tiles, x rows staged in LDS,
// weight columns read once per row-tile instead of once per row), so
// weight traffic drops ~16x and every phase has 16-64 workgroups of
// parallelism.  nsteps loop inside the kernel => zero host involvement
// between steps (the hipGraph path is only needed for the wide-batch
// configs that use the per-layer kernels above).
#define PNWG 96
// derived persistent-grid split points (fractions of PNWG)
#define PW4 (PNWG / 4)
#define PW3A (PNWG / 3)
#define PW3B (PNWG - 2 * (PNWG / 3))
// 4-row x 64-col tiles: a 64x256x256 GEMM phase becomes 64 tiles — the
// whole grid — with ~1 us of FMA per tile (16-row tiles measured ~6 us of
// per-wave compute and left 48 workgroups idle; scripts/tile_bench.hip)
#define PROWS 4
#define PXMAX 512
#define PWOFF (PROWS * PXMAX)
// pool sized for the chain variant's layout (2 activation row-buffers +
// 2 weight chunks), which is the largest user
#define PLDS_FLOATS (2 * 4 * PXMAX + 2 * 64 * 68 + 64)

struct PStepArgs {
    int B, O, A, H, K;
    float v_min, v_max, gamma_n, per_eps, tau, lr_actor, lr_critic;
    float per_alpha, per_beta0, per_beta_iters;
    int is_weighting;
    uint64_t seed;
    long tree_cap;
    long n_actor, n_critic;
    LayerDesc al[4], cl[4];
    // params / grads / moments / targets
    float *p_actor, *p_actor_t, *p_critic, *p_critic_t;
    float *g_actor, *g_critic, *m_actor, *v_actor, *m_critic, *v_critic;
    // replay + trees
    const float *rs, *ra, *rr, *rs2, *rd;
    double *sum_tree, *min_tree;
    // batch (bs is double-buffered: the policy-block tail pre-samples
    // the NEXT step's batch while this step's bs is still read by the
    // policy forward and the actor dW)
    float *bs, *bs_b, *ba, *br, *bs2, *bd, *bw, *pri;
    long *bidx;
    // activations / deltas workspace
    float *at_h1, *at_h2, *at_h3, *a2;
    float *ct_h1, *ct_h2, *ct_h3, *p_t, *m_proj;
    float *c_h1, *c_h2, *c_h3, *q, *dlog, *d3, *d2, *d1;
    float *pa_h1, *pa_h2, *pa_h3, *a_out;
    float *pc_h1, *pc_h2, *pc_h3, *pq;
    float *pd3, *pd2, *pdh1, *adz, *az1, *az2, *az3;
    Counters* cnt;
    unsigned long long* gbar;        // [0] = arrival counter, [1] = base
    unsigned long long* tstamp;      // [64] per-phase s_memrealtime stamps
};

// phase timing probe: wg0/lane0 stamps the wall clock (100 MHz constant
// clock) after each barrier of the FIRST step of a launch; read back via
// read_buffer("tstamp") to locate slow phases.
#define PTIME(g, s, i) \
    do { if (blockIdx.x == 0 && threadIdx.x == 0 && (s) == 0) \
        (g).tstamp[i] = __builtin_amdgcn_s_memrealtime(); } while (0)

// software grid barrier — tree arrival + write-once go-flag.  Safe because
// all PNWG workgroups are co-resident (256 wgs of 256 threads, one per CU
// — validated by the ctor's occupancy gate).
//
// Layout in gbar[]: 8 group counters at [g*16] (separate 128 B lines),
// root counter at [128], go-flag at [144], cross-launch round base at
// [160] (written by wg 0 after the final barrier; stream order makes the
// next launch's read race-free).
//
// Design notes (measured): a single central atomic that everyone arrives
// on AND polls makes idle workgroups' poll storm contend with the active
// workgroups' real memory traffic — single-GEMM phases went from ~6 us to
// ~40 us under it.  Here arrivals fan in through per-group lines, pollers
// read only the once-written flag, and the spin backs off exponentially
// (s_sleep needs an immediate, hence the two-stage backoff).
__device__ inline void p_bar(unsigned long long* gb,
                             unsigned long long& round) {
    __syncthreads();
    round += 1;
    if (threadIdx.x == 0) {
        __threadfence();                       // release: flush our writes
        int grp = blockIdx.x & 7;
        unsigned long long old = atomicAdd(&gb[grp << 4], 1ull);
        if (old + 1 == round * (PNWG >> 3)) {  // last of the group
            unsigned long long r = atomicAdd(&gb[128], 1ull);
            if (r + 1 == round * 8)            // last group overall
                __hip_atomic_store(&gb[144], round, __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
        }
        volatile unsigned long long* f = &gb[144];
        volatile unsigned long long* err = &gb[176];  // timeout flag (own line)
        long spins = 0;
        while (*f < round) {
            // safety valve: a barrier logic bug (or non-co-resident grid)
            // must never hard-hang the GPU.  On timeout (~10 s) raise the
            // error flag and bail; other workgroups see the flag on their
            // next poll-check and drain out instead of each spinning the
            // full 10 s.  The host checks the flag after every step() /
            // sync and throws — the step's results are NOT used.
            if (++spins > (1L << 26)) {
                atomicAdd((unsigned long long*)err, 1ull);
                break;
            }
            // barriers resolve in ~2-5 us: stay on the fast poll long
            // enough to catch that window before backing off.  The error
            // check rides the slow path only (every 64th backoff poll, to
            // keep poll traffic off the shared error line).
            if (spins < 48) __builtin_amdgcn_s_sleep(2);
            else {
                __builtin_amdgcn_s_sleep(32);
                if ((spins & 63) == 0 && *err) break;
            }
        }
        __threadfence();                       // acquire: invalidate L1
    }
    __syncthreads();
}

// quad-local barrier: synchronizes the 4 workgroups of one quad (a
// row-group's column-tile team in the chained fwd/bwd blocks) without
// touching the grid barrier.  Each quad owns its own 128 B line at
// gbar[192 + q*16]; arrivals and polls stay quad-private, so 16 quads
// proceed fully independently (~0.5-1 us vs ~3.3 us for the grid form).
// qround carries across launches like the grid round: members read the
// line once at kernel entry (ordered before any arrival by the first
// grid barrier) and count from there.
__device__ inline void q_bar(unsigned long long* qc,
                             unsigned long long& qround) {
    __syncthreads();
    qround += 4;
    if (threadIdx.x == 0) {
        __threadfence();
        atomicAdd(qc, 1ull);
        volatile unsigned long long* f = qc;
        long spins = 0;
        while (*f < qround) {
            if (++spins > (1L << 26)) break;
            __builtin_amdgcn_s_sleep(1);
        }
        __threadfence();
    }
    __syncthreads();
}
// (A per-member slot-store variant — release stores + one-line relaxed
// polls instead of the shared atomic counter — measured SLOWER: 3674 vs
// 3773 steps/s; the four poll loads don't coalesce into one L2 access.)

// group barrier for the policy-block tail crew (q_bar with a caller-set
// member count; own line at gbar[1280])
__device__ inline void t_bar(unsigned long long* tc,
                             unsigned long long& trnd, int members) {
    __syncthreads();
    trnd += members;
    if (threadIdx.x == 0) {
        __threadfence();
        atomicAdd(tc, 1ull);
        volatile unsigned long long* f = tc;
        long spins = 0;
        while (*f < trnd) {
            if (++spins > (1L << 26)) break;
            __builtin_amdgcn_s_sleep(1);
        }
        __threadfence();
    }
    __syncthreads();
}

// Tiled forward: 16-row x 64-col tiles; x rows staged in LDS (broadcast
// reads), weight column read ONCE per tile (not once per row).  Thread
// (rq = tid/64, c = tid%64) accumulates rows {r0+rq, +4, +8, +12} of column
// c0+c — four independent FMA chains for ILP; softmax (out<=64) reduces
// per-row across the wave.  Summation order over k is ascending, matching
// the eager/rowblock paths bit-for-bit.
// (__noinline__ is load-bearing: inlining 15 copies into the persistent
// kernel segfaults ROCm 7.2's clang in ADCE; the call overhead is noise
// next to the ~1 us phase time.)
__device__ __noinline__ void p_fwd(float* lds, const float* x1, const float* x2,
                             const float* wt, const float* bias, float* y,
                             int B, int in1, int in2, int out, int act_kind,
                             int wg_rel, int nwg) {
    int in_total = in1 + in2;
    // x rows staged at a 4-float-aligned stride so the k-loop can read
    // them with ds_read_b128 (16 B alignment)
    int in_pad = (in_total + 3) & ~3;
    int nrt = (B + PROWS - 1) / PROWS;
    int nct = (out + 63) / 64;
    int ntiles = nrt * nct;
    int tid = threadIdx.x;
    for (int t = wg_rel; t < ntiles; t += nwg) {
        int r0 = (t / nct) * PROWS, c0 = (t % nct) * 64;
        int kk16 = tid >> 6, cc16 = tid & 63;
        float wreg[16];
        int nfull = in_total >> 6;
        auto preload = [&](int kc) {
#pragma unroll
            for (int u = 0; u < 16; ++u) {
                int gk = kc + kk16 + 4 * u, gc = c0 + cc16;
                wreg[u] = (gc < out)
                    ? wt[(long)gk * out + gc] : 0.f;
            }
        };
        // first weight chunk's loads fly WITH the x-stage loads (the
        // x ds_writes only wait on the older x loads)
        if (nfull > 0) preload(0);
        // x-stage, register-batched (all loads in flight, then ds_writes)
        {
            int tot = PROWS * in_pad;
            for (int base = 0; base < tot; base += 256 * 8) {
                float tmp[8];
#pragma unroll
                for (int u = 0; u < 8; ++u) {
                    int e = base + u * 256 + tid;
                    float v = 0.f;
                    if (e < tot) {
                        int rr_ = e / in_pad, kk = e % in_pad;
                        int gb = r0 + rr_;
                        if (gb < B && kk < in_total)
                            v = (kk < in1)
                                ? x1[(long)gb * in1 + kk]
                                : x2[(long)gb * in2 + (kk - in1)];
                    }
                    tmp[u] = v;
                }
#pragma unroll
                for (int u = 0; u < 8; ++u) {
                    int e = base + u * 256 + tid;
                    if (e < tot) lds[e] = tmp[u];
                }
            }
        }
        __syncthreads();
        int rq = tid >> 6, c = tid & 63;
        int o = c0 + c;
        int r = r0 + rq;
        float acc = 0.f;
        // weight slice K-chunked through LDS in [col][k] layout (row
        // stride 68 floats: 16 B-aligned and 4-bank-strided across
        // lanes), so the k-loop fetches 4 k at a time with ONE
        // ds_read_b128 per operand — the b32 form issued 3 LDS reads
        // per k and was issue-bound, not bandwidth-bound.  Summation
        // stays ascending-k: bitwise identical to the eager oracle.
        float* ws = lds + PWOFF;                 // [64 cols][68 k]
        const float* xr = lds + rq * in_pad;
        for (int ch = 0; ch < nfull; ++ch) {
            int kc = ch << 6;
            float wb[16];
#pragma unroll
            for (int u = 0; u < 16; ++u) wb[u] = wreg[u];
            if (kc + 64 < (nfull << 6)) preload(kc + 64);
#pragma unroll
            for (int u = 0; u < 16; ++u)
                ws[cc16 * 68 + kk16 + 4 * u] = wb[u];
            __syncthreads();
            if (o < out) {
                const float* wcol = ws + c * 68;
#pragma unroll
                for (int k = 0; k < 64; k += 4) {
                    float4 wv = *reinterpret_cast<const float4*>(wcol + k);
                    float4 xv = *reinterpret_cast<const float4*>(
                        xr + kc + k);
                    acc += xv.x * wv.x;
                    acc += xv.y * wv.y;
                    acc += xv.z * wv.z;
                    acc += xv.w * wv.w;
                }
            }
            __syncthreads();
        }
        int kc = nfull << 6, klen = in_total - kc;
        if (klen > 0) {
            for (int e = tid; e < (klen << 6); e += 256) {
                int kk = e >> 6, cc = e & 63;
                int gc = c0 + cc;
                ws[cc * 68 + kk] = (gc < out)
                    ? wt[(long)(kc + kk) * out + gc] : 0.f;
            }
            __syncthreads();
            if (o < out) {
                for (int k = 0; k < klen; ++k)
                    acc += xr[kc + k] * ws[c * 68 + k];
            }
            __syncthreads();
        }
        if (act_kind == ACT_SOFTMAX) {
            // out <= 64, single col-chunk; wave == one row
            float v = (o < out) ? acc + bias[o] : -INFINITY;
            float mx = v;
            for (int s = 32; s > 0; s >>= 1)
                mx = fmaxf(mx, __shfl_xor(mx, s, 64));
            float e = (o < out) ? __expf(v - mx) : 0.f;
            float sum = e;
            for (int s = 32; s > 0; s >>= 1) sum += __shfl_xor(sum, s, 64);
            if (r < B && o < out) y[(long)r * out + o] = e / sum;
        } else if (r < B && o < out) {
            float v = acc + bias[o];
            if (act_kind == ACT_RELU) v = fmaxf(v, 0.f);
            else if (act_kind == ACT_TANH) v = tanhf(v);
            y[(long)r * out + o] = v;
        }
        __syncthreads();
    }
}

// Tiled backward-dX: dx[b][i-in_lo] = (sum_o dz[b][o] wt[i][o]) * mask.
// dz rows staged in LDS; mask/dx arrays are [B][span] (span = in_hi-in_lo):
// full-range h arrays and the concat action slice both fit this form.
__device__ inline void p_bwd_dx(float* lds, const float* dz, const float* wt,
                                int in_lo, int in_hi, int out, int B,
                                const float* hprev, int prev_act, float* dx,
                                int wg_rel, int nwg) {
    int span = in_hi - in_lo;
    int nrt = (B + PROWS - 1) / PROWS;
    int nit = (span + 63) / 64;
    int ntiles = nrt * nit;
    int tid = threadIdx.x;
    for (int t = wg_rel; t < ntiles; t += nwg) {
        int r0 = (t / nit) * PROWS, i0 = in_lo + (t % nit) * 64;
        int oo16 = tid & 3;
        int ii16 = tid >> 2;
        float wreg[16];
        int gi_ld = i0 + ii16;
        int nfull = out >> 6;
        auto preload = [&](int oc) {
#pragma unroll
            for (int u = 0; u < 16; ++u) {
                int go = oc + oo16 + 4 * u;
                wreg[u] = (gi_ld < in_hi)
                    ? wt[(long)gi_ld * out + go] : 0.f;
            }
        };
        if (nfull > 0) preload(0);
        // dz-stage, register-batched; rows padded to 4-float stride so
        // the o-loop reads them with ds_read_b128
        int out_pad = (out + 3) & ~3;
        {
            int tot = PROWS * out_pad;
            for (int base = 0; base < tot; base += 256 * 8) {
                float tmp[8];
#pragma unroll
                for (int u = 0; u < 8; ++u) {
                    int e = base + u * 256 + tid;
                    float v = 0.f;
                    if (e < tot) {
                        int rr_ = e / out_pad, oo = e % out_pad;
                        int gb = r0 + rr_;
                        if (gb < B && oo < out)
                            v = dz[(long)gb * out + oo];
                    }
                    tmp[u] = v;
                }
#pragma unroll
                for (int u = 0; u < 8; ++u) {
                    int e = base + u * 256 + tid;
                    if (e < tot) lds[e] = tmp[u];
                }
            }
        }
        __syncthreads();
        int rq = tid >> 6, c = tid & 63;
        int i = i0 + c;
        int r = r0 + rq;
        float acc = 0.f;
        // weight tile O-chunked through LDS in [i][o] layout (stride 68:
        // 16 B-aligned rows), so the o-loop fetches 4 o per ds_read_b128
        // for both operands — see p_fwd's identical rework; summation
        // stays ascending-o, bitwise identical
        float* ws = lds + PWOFF;                 // [64 i][68 o]
        const float* zr = lds + rq * out_pad;
        for (int ch = 0; ch < nfull; ++ch) {
            int oc = ch << 6;
            float wb[16];
#pragma unroll
            for (int u = 0; u < 16; ++u) wb[u] = wreg[u];
            if (oc + 64 < (nfull << 6)) preload(oc + 64);
#pragma unroll
            for (int u = 0; u < 16; ++u)
                ws[ii16 * 68 + oo16 + 4 * u] = wb[u];
            __syncthreads();
            if (i < in_hi) {
                const float* wrow = ws + c * 68;
#pragma unroll
                for (int o = 0; o < 64; o += 4) {
                    float4 wv = *reinterpret_cast<const float4*>(wrow + o);
                    float4 zv = *reinterpret_cast<const float4*>(
                        zr + oc + o);
                    acc += zv.x * wv.x;
                    acc += zv.y * wv.y;
                    acc += zv.z * wv.z;
                    acc += zv.w * wv.w;
                }
            }
            __syncthreads();
        }
        int oc = nfull << 6, olen = out - oc;
        if (olen > 0) {
            for (int e = tid; e < 4096; e += 256) {
                int ii = e >> 6, oo = e & 63;
                int gi = i0 + ii;
                ws[ii * 68 + oo] = (gi < in_hi && oo < olen)
                    ? wt[(long)gi * out + (oc + oo)] : 0.f;
            }
            __syncthreads();
            if (i < in_hi) {
                for (int o = 0; o < olen; ++o)
                    acc += zr[oc + o] * ws[c * 68 + o];
            }
            __syncthreads();
        }
        if (r < B && i < in_hi) {
            long rel = (long)r * span + (i - in_lo);
            float m_ = hprev ? act_mask(prev_act, hprev[rel]) : 1.f;
            dx[rel] = acc * m_;
        }
        __syncthreads();
    }
}

// Tiled dW (+db on the i0==0 tile row): 16x16 tiles, B-chunked LDS staging,
// identical math/order to bwd_one's dW part.
__device__ inline void p_dw(float* lds, const float* dz, const float* x1,
                            const float* x2, float* dwt, float* dbias,
                            int B, int in1, int in2, int out,
                            int wg_rel, int nwg) {
    int in_total = in1 + in2;
    int nti = (in_total + BWT - 1) / BWT, nto = (out + BWT - 1) / BWT;
    int ntiles = nti * nto;
    float* xs = lds;                       // [64][BWT+1]
    float* zs = lds + 64 * (BWT + 1);      // [64][BWT+1]
    int tid = threadIdx.x;
    int ti = tid / BWT, to = tid % BWT;
    for (int t = wg_rel; t < ntiles; t += nwg) {
        int i0 = (t / nto) * BWT, o0 = (t % nto) * BWT;
        float acc = 0.f, accb = 0.f;
        for (int bc = 0; bc < B; bc += 64) {
            for (int e = tid; e < 64 * BWT; e += 256) {
                int bb = e / BWT, ii = e % BWT;
                int gb = bc + bb, gi = i0 + ii;
                float xv = 0.f;
                if (gb < B && gi < in_total)
                    xv = (gi < in1) ? x1[(long)gb * in1 + gi]
                                    : x2[(long)gb * in2 + (gi - in1)];
                xs[bb * (BWT + 1) + ii] = xv;
                int go = o0 + ii;
                zs[bb * (BWT + 1) + ii] =
                    (gb < B && go < out) ? dz[(long)gb * out + go] : 0.f;
            }
            __syncthreads();
#pragma unroll 16
            for (int bb = 0; bb < 64; ++bb) {
                acc += xs[bb * (BWT + 1) + ti] * zs[bb * (BWT + 1) + to];
                if (ti == 0) accb += zs[bb * (BWT + 1) + to];
            }
            __syncthreads();
        }
        int gi = i0 + ti, go = o0 + to;
        if (gi < in_total && go < out) dwt[(long)gi * out + go] = acc;
        if (ti == 0 && go < out && i0 == 0 && dbias) dbias[go] = accb;
    }
}

// dW tile [64 i x 64 o] per wg: x^T and dz chunks staged in LDS (coalesced),
// each thread owns a 4x4 output patch (ig = tid>>4 picks 4 i's, og = tid&15
// picks 4 o's) — 8 LDS reads + 16 FMAs per batch row.  Bias rows fold in on
// the i0==0 tiles.  Accumulation over b ascending, same order as p_dw/bwd_one.
__device__ __noinline__ void p_dw2(float* lds, const float* dz,
                                   const float* x1, const float* x2,
                                   float* dwt, float* dbias,
                                   int B, int in1, int in2, int out,
                                   int wg_rel, int nwg) {
    int in_total = in1 + in2;
    int nti = (in_total + 63) >> 6, nto = (out + 63) >> 6;
    int ntiles = nti * nto;
    float* xs = lds;                       // [64 b][68] i-slice
    float* zs = lds + 64 * 68;             // [64 b][68] o-slice
    // row stride 68: 16 B-aligned so the b-loop reads each 4-wide
    // register block with ONE ds_read_b128 (see p_fwd's rework)
    int tid = threadIdx.x;
    int ig = tid >> 4, og = tid & 15;
    for (int t = wg_rel; t < ntiles; t += nwg) {
        int i0 = (t / nto) << 6, o0 = (t % nto) << 6;
        float acc[4][4] = {};
        float accb[4] = {};
        for (int bc = 0; bc < B; bc += 64) {
            // register-batched staging (all 32 loads in flight)
            float tx[16], tz[16];
#pragma unroll
            for (int u = 0; u < 16; ++u) {
                int e = u * 256 + tid;
                int bb = e >> 6, ii = e & 63;
                int gb = bc + bb, gi = i0 + ii;
                float xv = 0.f;
                if (gb < B && gi < in_total)
                    xv = (gi < in1) ? x1[(long)gb * in1 + gi]
                                    : x2[(long)gb * in2 + (gi - in1)];
                tx[u] = xv;
                int go = o0 + ii;
                tz[u] = (gb < B && go < out)
                    ? dz[(long)gb * out + go] : 0.f;
            }
#pragma unroll
            for (int u = 0; u < 16; ++u) {
                int e = u * 256 + tid;
                int bb = e >> 6, ii = e & 63;
                xs[bb * 68 + ii] = tx[u];
                zs[bb * 68 + ii] = tz[u];
            }
            __syncthreads();
            int blim = min(64, B - bc);
            for (int b = 0; b < blim; ++b) {
                float4 xq = *reinterpret_cast<const float4*>(
                    xs + b * 68 + ig * 4);
                float4 zq = *reinterpret_cast<const float4*>(
                    zs + b * 68 + og * 4);
                float xv0 = xq.x, xv1 = xq.y, xv2 = xq.z, xv3 = xq.w;
                float zv0 = zq.x, zv1 = zq.y, zv2 = zq.z, zv3 = zq.w;
                acc[0][0] += xv0 * zv0; acc[0][1] += xv0 * zv1;
                acc[0][2] += xv0 * zv2; acc[0][3] += xv0 * zv3;
                acc[1][0] += xv1 * zv0; acc[1][1] += xv1 * zv1;
                acc[1][2] += xv1 * zv2; acc[1][3] += xv1 * zv3;
                acc[2][0] += xv2 * zv0; acc[2][1] += xv2 * zv1;
                acc[2][2] += xv2 * zv2; acc[2][3] += xv2 * zv3;
                acc[3][0] += xv3 * zv0; acc[3][1] += xv3 * zv1;
                acc[3][2] += xv3 * zv2; acc[3][3] += xv3 * zv3;
                if (ig == 0 && i0 == 0) {
                    accb[0] += zv0; accb[1] += zv1;
                    accb[2] += zv2; accb[3] += zv3;
                }
            }
            __syncthreads();
        }
#pragma unroll
        for (int a = 0; a < 4; ++a) {
            int gi = i0 + ig * 4 + a;
            if (gi >= in_total) continue;
#pragma unroll
            for (int bb = 0; bb < 4; ++bb) {
                int go = o0 + og * 4 + bb;
                if (go < out) dwt[(long)gi * out + go] = acc[a][bb];
            }
        }
        if (ig == 0 && i0 == 0 && dbias) {
#pragma unroll
            for (int bb = 0; bb < 4; ++bb) {
                int go = o0 + og * 4 + bb;
                if (go < out) dbias[go] = accb[bb];
            }
        }
    }
}

__device__ inline void p_adam_lerp(float* __restrict__ p,
                                   const float* __restrict__ gr,
                                   float* __restrict__ m,
                                   float* __restrict__ v,
                                   float* __restrict__ tgt_slab, long n,
                                   float lr, float tau, long long t,
                                   int nwg = PNWG) {
    const float b1 = 0.9f, b2 = 0.999f, eps = 1e-8f;
    float bc1 = 1.f - __powf(b1, (float)t);
    float bc2 = 1.f - __powf(b2, (float)t);
    // register-batched: fire all 5 streams' loads for 4 grid-strides at
    // once (a serial load->compute->store loop pays one memory round-trip
    // per element batch)
    long stride = (long)nwg * 256;
    long i0 = (long)blockIdx.x * 256 + threadIdx.x;
    for (long base = i0; base < n; base += stride * 4) {
        float gi_[4], mi_[4], vi_[4], pi_[4], ti_[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
            long i = base + u * stride;
            bool ok = i < n;
            gi_[u] = ok ? gr[i] : 0.f;
            mi_[u] = ok ? m[i] : 0.f;
            vi_[u] = ok ? v[i] : 0.f;
            pi_[u] = ok ? p[i] : 0.f;
            ti_[u] = ok ? tgt_slab[i] : 0.f;
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
            long i = base + u * stride;
            if (i >= n) continue;
            float gi = gi_[u];
            float mi = b1 * mi_[u] + (1.f - b1) * gi;
            float vi = b2 * vi_[u] + (1.f - b2) * gi * gi;
            m[i] = mi; v[i] = vi;
            float pn = pi_[u] - lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
            p[i] = pn;
            tgt_slab[i] = ti_[u] + tau * (pn - ti_[u]);
        }
    }
}

// PER sample + gather phase (k_per_sample semantics; one wave per probe)
__device__ inline void p_sample(const PStepArgs& g, int wg0 = 0,
                                int sched_off = 0,
                                long long epoch_ovr = -1,
                                long long beta_ovr = -1,
                                float* bs_out = nullptr) {
    // bs_out: which state buffer to gather into (double-buffered in the
    // persistent kernel; defaults to g.bs)
    // sched_off = +1 when pre-sampling the NEXT step's batch before the
    // counter tick (overlapped under the actor-Adam phase).  epoch_ovr /
    // beta_ovr: explicit schedule values from the hoisted-counter path
    // (k_step_persistent reads the counters ONCE per launch and threads
    // base+s through; loss zeroing then lives in the ct.L3 phase).
    int probe = ((int)blockIdx.x - wg0) * 4 + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    if (epoch_ovr < 0 && sched_off == 0 && probe == 0 && lane == 0) {
        g.cnt->loss_critic = 0.f;
        g.cnt->loss_actor = 0.f;
    }
    if (probe >= g.B) return;
    long long n = g.cnt->size;
    double total = g.sum_tree[1];
    long long beta_t = beta_ovr >= 0 ? beta_ovr
                                     : g.cnt->beta_t + sched_off;
    float frac = fminf((float)((double)beta_t / g.per_beta_iters), 1.0f);
    float beta = g.per_beta0 + frac * (1.0f - g.per_beta0);
    long idx;
    if (lane == 0) {
        long long epoch = epoch_ovr >= 0 ? epoch_ovr
                                         : g.cnt->rng_epoch + sched_off;
        Philox4 r = philox4(g.seed, (uint64_t)epoch, (uint64_t)probe);
        double mass = (double)u01(r.v[0]) * total;
        long node = 1;
        // 4-ary descent over the binary layout (see k_per_sample): two
        // binary levels per contiguous grandchildren load, bit-exact leaf
        // choice.  Deep levels are random-access, once-per-step: bypass
        // L2 for them, keep the hot top of the tree cached.
        while (node < g.tree_cap / 2) {
            long gb = 4 * node;
            double g0, g1, g2;
            if (gb >= 16384) {
                g0 = __builtin_nontemporal_load(&g.sum_tree[gb]);
                g1 = __builtin_nontemporal_load(&g.sum_tree[gb + 1]);
                g2 = __builtin_nontemporal_load(&g.sum_tree[gb + 2]);
            } else {
                g0 = g.sum_tree[gb];
                g1 = g.sum_tree[gb + 1];
                g2 = g.sum_tree[gb + 2];
            }
            double s01 = g0 + g1;
            if (mass <= s01) {
                if (mass <= g0) node = gb;
                else { mass -= g0; node = gb + 1; }
            } else {
                mass -= s01;
                if (mass <= g2) node = gb + 2;
                else { mass -= g2; node = gb + 3; }
            }
        }
        if (node < g.tree_cap) {               // odd final level
            double ls = (node >= 16384)
                ? __builtin_nontemporal_load(&g.sum_tree[2 * node])
                : g.sum_tree[2 * node];
            if (mass > ls) { mass -= ls; node = 2 * node + 1; }
            else           { node = 2 * node; }
        }
        idx = node - g.tree_cap;
        if (idx >= n) idx = n - 1;
        g.bidx[probe] = idx;
        double p = g.sum_tree[g.tree_cap + idx] / total;
        double p_min = g.min_tree[1] / total;
        double max_w = pow(p_min * (double)n, (double)-beta);
        g.bw[probe] = (float)(pow(p * (double)n, (double)-beta) / max_w);
        g.br[probe] = g.rr[idx];
        g.bd[probe] = g.rd[idx];
    }
    idx = __shfl(idx, 0, 64);
    float* bs_dst = bs_out ? bs_out : g.bs;
    // replay rows are random and never re-read — keep them out of L2
    // (the weights need it; SURVEY hot-loop note)
    for (int k = lane; k < g.O; k += 64) {
        bs_dst[(long)probe * g.O + k] =
            __builtin_nontemporal_load(&g.rs[idx * g.O + k]);
        g.bs2[(long)probe * g.O + k] =
            __builtin_nontemporal_load(&g.rs2[idx * g.O + k]);
    }
    for (int k = lane; k < g.A; k += 64)
        g.ba[(long)probe * g.A + k] =
            __builtin_nontemporal_load(&g.ra[idx * g.A + k]);
}

// C51 projection phase (k_project semantics; one wave per row, LDS row)
__device__ inline void p_project(const PStepArgs& g, float* lds) {
    int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
    int row = blockIdx.x * 4 + wid;
    int K = g.K;
    float* mrow = lds + wid * 64;
    for (int k = lane; k < K; k += 64) mrow[k] = 0.f;
    __builtin_amdgcn_wave_barrier();
    if (row < g.B && lane < K) {
        float delta = (g.v_max - g.v_min) / (K - 1);
        float z = g.v_min + lane * delta;
        float tz = g.br[row] + g.gamma_n * (1.f - g.bd[row]) * z;
        tz = fminf(g.v_max, fmaxf(g.v_min, tz));
        float b = (tz - g.v_min) / delta;
        int l = (int)floorf(b), u = (int)ceilf(b);
        if (l == u) { if (u > 0) l -= 1; else u += 1; }
        float p = g.p_t[(long)row * K + lane];
        atomicAdd(&mrow[l], p * ((float)u - b));
        atomicAdd(&mrow[u], p * (b - (float)l));
    }
    __builtin_amdgcn_wave_barrier();
    if (row < g.B)
        for (int k = lane; k < K; k += 64)
            g.m_proj[(long)row * K + k] = mrow[k];
}

// merged projection + CE-grad phase: the wave that projects row b's target
// distribution keeps it in LDS and computes the CE gradient/priority
// immediately (saves a barrier and the m_proj global round-trip; q comes
// from PH4).  m_proj is still written out for introspection/parity tests.
__device__ inline void p_project_ce(const PStepArgs& g, float* lds,
                                    int row_base = -1) {
    int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
    int row = (row_base < 0 ? (int)blockIdx.x * 4 : row_base) + wid;
    int K = g.K;
    float* mrow = lds + wid * 64;
    for (int k = lane; k < K; k += 64) mrow[k] = 0.f;
    __builtin_amdgcn_wave_barrier();
    if (row < g.B && lane < K) {
        float delta = (g.v_max - g.v_min) / (K - 1);
        float z = g.v_min + lane * delta;
        float tz = g.br[row] + g.gamma_n * (1.f - g.bd[row]) * z;
        tz = fminf(g.v_max, fmaxf(g.v_min, tz));
        float b = (tz - g.v_min) / delta;
        int l = (int)floorf(b), u = (int)ceilf(b);
        if (l == u) { if (u > 0) l -= 1; else u += 1; }
        float p = g.p_t[(long)row * K + lane];
        atomicAdd(&mrow[l], p * ((float)u - b));
        atomicAdd(&mrow[u], p * (b - (float)l));
    }
    __builtin_amdgcn_wave_barrier();
    if (row >= g.B) return;
    float qv = 0.f, mv = 0.f;
    if (lane < K) {
        g.m_proj[(long)row * K + lane] = mrow[lane];
        qv = g.q[(long)row * K + lane];
        mv = mrow[lane];
    }
    float dot = mv * qv;
    float ce = -mv * __logf(qv + 1e-10f);
    for (int s = 32; s > 0; s >>= 1) {
        dot += __shfl_xor(dot, s, 64);
        ce += __shfl_xor(ce, s, 64);
    }
    float scale = (g.is_weighting && g.bw) ? g.bw[row] : 1.f;
    if (lane < K)
        g.dlog[(long)row * K + lane] = scale * (qv - mv) / (float)g.B;
    if (lane == 0) {
        g.pri[row] = dot + g.per_eps;
        atomicAdd(&g.cnt->loss_critic, scale * ce / (float)g.B);
    }
}

// narrow backward-dX (span <= 8, e.g. the concat action slice): one wave
// per batch row, lane-parallel dot over `out` with a shfl reduce, tanh
// mask applied from ymask.
__device__ inline void p_bwd_dx_narrow(const PStepArgs& g, const float* dz,
                                       const float* wt, int in_lo,
                                       int in_hi, int out,
                                       const float* ymask, float* dx,
                                       int row_base = -1) {
    int row = (row_base < 0 ? (int)blockIdx.x * 4 : row_base)
              + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    if (row >= g.B) return;
    int span = in_hi - in_lo;
    for (int j = 0; j < span; ++j) {
        const float* wrow = wt + (long)(in_lo + j) * out;
        float acc = 0.f;
        for (int o = lane; o < out; o += 64)
            acc += dz[(long)row * out + o] * wrow[o];
        for (int s = 32; s > 0; s >>= 1) acc += __shfl_xor(acc, s, 64);
        if (lane == 0) {
            float y = ymask[(long)row * span + j];
            dx[(long)row * span + j] = acc * (1.f - y * y);
        }
    }
}

__device__ inline void p_ce_grad(const PStepArgs& g) {
    int row = blockIdx.x * 4 + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    if (row >= g.B) return;
    int K = g.K;
    float qv = 0.f, mv = 0.f;
    if (lane < K) {
        qv = g.q[(long)row * K + lane];
        mv = g.m_proj[(long)row * K + lane];
    }
    float dot = mv * qv;
    float ce = -mv * __logf(qv + 1e-10f);
    for (int s = 32; s > 0; s >>= 1) {
        dot += __shfl_xor(dot, s, 64);
        ce += __shfl_xor(ce, s, 64);
    }
    float scale = (g.is_weighting && g.bw) ? g.bw[row] : 1.f;
    if (lane < K)
        g.dlog[(long)row * K + lane] = scale * (qv - mv) / (float)g.B;
    if (lane == 0) {
        g.pri[row] = dot + g.per_eps;
        atomicAdd(&g.cnt->loss_critic, scale * ce / (float)g.B);
    }
}

__device__ inline void p_policy_grad(const PStepArgs& g,
                                     int row_base = -1) {
    int row = (row_base < 0 ? (int)blockIdx.x * 4 : row_base)
              + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    if (row >= g.B) return;
    int K = g.K;
    float delta = (g.v_max - g.v_min) / (K - 1);
    float z = g.v_min + lane * delta;
    float qv = (lane < K) ? g.pq[(long)row * K + lane] : 0.f;
    float e = qv * z;
    for (int s = 32; s > 0; s >>= 1) e += __shfl_xor(e, s, 64);
    if (lane < K)
        g.pd3[(long)row * K + lane] = -qv * (z - e) / (float)g.B;
    if (lane == 0) atomicAdd(&g.cnt->loss_actor, -e / (float)g.B);
}

// PER write-back + schedule-counter tick (k_per_update semantics; wg 0 only)
__device__ inline void p_per_update(const PStepArgs& g, bool do_tick = true,
                                    int owner_wg = 0) {
    if ((int)blockIdx.x != owner_wg) return;
    int tid = threadIdx.x;
    float local_max = 0.f;
    for (int i = tid; i < g.B; i += 256) {
        float p = g.pri[i];
        double pa = pow((double)p, (double)g.per_alpha);
        long leaf = g.tree_cap + g.bidx[i];
        g.sum_tree[leaf] = pa;
        g.min_tree[leaf] = pa;
        local_max = fmaxf(local_max, p);
    }
    __shared__ float smax[256];
    smax[tid] = local_max;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {
        if (tid < s) smax[tid] = fmaxf(smax[tid], smax[tid + s]);
        __syncthreads();
    }
    if (tid == 0)
        g.cnt->max_priority = fmaxf(g.cnt->max_priority, smax[0]);
    __syncthreads();
    long levels = 0;
    for (long c = g.tree_cap; c > 1; c >>= 1) ++levels;
    for (long lv = 0; lv < levels; ++lv) {
        for (int i = tid; i < g.B; i += 256) {
            long node = (g.tree_cap + g.bidx[i]) >> (lv + 1);
            if (node >= 1) {
                g.sum_tree[node] =
                    g.sum_tree[2 * node] + g.sum_tree[2 * node + 1];
                g.min_tree[node] = fmin(g.min_tree[2 * node],
                                        g.min_tree[2 * node + 1]);
            }
        }
        __syncthreads();
    }
    if (do_tick && tid == 0) {
        g.cnt->beta_t += 1;
        g.cnt->adam_t_actor += 1;
        g.cnt->adam_t_critic += 1;
        g.cnt->rng_epoch += 1;
    }
}

// ===========================================================================
// Chain-fused persistent step (flagship path v2)
// ===========================================================================
// Forward layers and backward-dX layers are ROW-LOCAL: y[b] depends only on
// x[b].  So a workgroup that owns PR complete rows chains through an entire
// network (and back) with only __syncthreads — no grid barriers between
// layers.  The step collapses to ~10 grid-barrier phases:
//   sample | {actor_t, critic, actor} fwd chains | critic_t chain+proj+CE |
//   critic dX chain | critic dW | critic Adam | policy megachain (fwd +
//   policy grad + dX back through critic AND actor, with the PER tree
//   write-back overlapped on a spare workgroup) | actor dW | actor Adam |
//   tick.
// Activations stay in LDS between layers ([PR][PXMAX] ping-pong rows);
// weights stream through double-buffered 64x64 LDS chunks with register-
// batched preloads (same staging discipline as p_fwd).
#define PR 4
// LDS pool reuse during chain phases: XA | XB | WS0 | WS1
#define PX_A 0
#define PX_B (PR * PXMAX)
#define PW_0 (2 * PR * PXMAX)
#define PW_1 (2 * PR * PXMAX + 64 * 65)

// stage PR rows of src[B][width] into x_l rows at column offset `off`,
// zero-padding columns [off+width, pad_to)
__device__ inline void p_rows_load(float* x_l, const float* src, long r0,
                                   int B, int width, int off, int pad_to) {
    int tid = threadIdx.x;
    int tot = PR * width;
    for (int base = 0; base < tot; base += 256 * 8) {
        float tmp[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) {
            int e = base + u * 256 + tid;
            float v = 0.f;
            if (e < tot) {
                int rr = e / width, kk = e % width;
                if (r0 + rr < B) v = src[(r0 + rr) * width + kk];
            }
            tmp[u] = v;
        }
#pragma unroll
        for (int u = 0; u < 8; ++u) {
            int e = base + u * 256 + tid;
            if (e < tot)
                x_l[(e / width) * PXMAX + off + (e % width)] = tmp[u];
        }
    }
    for (int e = tid; e < PR * (pad_to - off - width); e += 256) {
        int span = pad_to - off - width;
        x_l[(e / span) * PXMAX + off + width + (e % span)] = 0.f;
    }
}

// one dense layer for a PR-row group: y_l = act(x_l @ W + b), LDS-resident;
// optionally mirrors y to global ysave[B][out].  Columns beyond `out` are
// zero-padded to the next 64 multiple so the next layer can run full-64
// K chunks.
__device__ __noinline__ void p_layer_rows(float* lds, const float* x_l,
                                          float* y_l, const float* wt,
                                          const float* bias, int in_total,
                                          int out, int act_kind,
                                          float* ysave, long r0, int B) {
    int tid = threadIdx.x, rq = tid >> 6, c = tid & 63;
    int kk16 = tid >> 6, cc16 = tid & 63;
    int nk = (in_total + 63) >> 6;
    int nc = (out + 63) >> 6;
    int total_ch = nc * nk;
    float* ws0 = lds + PW_0;
    float* ws1 = lds + PW_1;
    float wreg[16];
    auto preload = [&](int t) {
        int c0 = (t / nk) << 6, kc = (t % nk) << 6;
#pragma unroll
        for (int u = 0; u < 16; ++u) {
            int gk = kc + kk16 + 4 * u, gc = c0 + cc16;
            wreg[u] = (gk < in_total && gc < out)
                ? wt[(long)gk * out + gc] : 0.f;
        }
    };
    preload(0);
    float acc = 0.f;
    const float* xr = x_l + rq * PXMAX;
    for (int t = 0; t < total_ch; ++t) {
        int ci = t / nk, ki = t % nk;
        float* ws = (t & 1) ? ws1 : ws0;
        float wb[16];
#pragma unroll
        for (int u = 0; u < 16; ++u) wb[u] = wreg[u];
        if (t + 1 < total_ch) preload(t + 1);
#pragma unroll
        for (int u = 0; u < 16; ++u)
            ws[(kk16 + 4 * u) * 65 + cc16] = wb[u];
        __syncthreads();
        if (ki == 0) acc = 0.f;
        int kc = ki << 6;
#pragma unroll 8
        for (int k = 0; k < 64; ++k)
            acc += xr[kc + k] * ws[k * 65 + c];
        if (ki == nk - 1) {
            int o = (ci << 6) + c;
            float v = acc;
            bool live = (o < out) && (r0 + rq < B);
            if (o < out) {
                v += bias[o];
                if (act_kind == ACT_RELU) v = fmaxf(v, 0.f);
                else if (act_kind == ACT_TANH) v = tanhf(v);
            }
            if (act_kind == ACT_SOFTMAX) {
                // out <= 64 => single col chunk; wave == one row
                float lv = (o < out) ? v : -INFINITY;
                float mx = lv;
                for (int s = 32; s > 0; s >>= 1)
                    mx = fmaxf(mx, __shfl_xor(mx, s, 64));
                float e = (o < out) ? __expf(lv - mx) : 0.f;
                float sum = e;
                for (int s = 32; s > 0; s >>= 1)
                    sum += __shfl_xor(sum, s, 64);
                v = (sum > 0.f) ? e / sum : 0.f;
            }
            int pad_to = nc << 6;
            if (o < pad_to)
                y_l[rq * PXMAX + o] = (o < out) ? v : 0.f;
            if (live && ysave) ysave[(r0 + rq) * out + o] = v;
        }
        __syncthreads();
    }
}

// backward-dX for a PR-row group: dx_l = (dz_l @ W^T) * act'(h), LDS ->
// LDS, h/dx mirrors read/written against global [B][span] arrays.
__device__ __noinline__ void p_dx_rows(float* lds, const float* dz_l,
                                       float* dx_l, const float* wt,
                                       int in_lo, int in_hi, int out,
                                       const float* hsave, int prev_act,
                                       float* dxsave, long r0, int B) {
    int tid = threadIdx.x, rq = tid >> 6, c = tid & 63;
    int oo16 = tid & 3, ii16 = tid >> 2;
    int span = in_hi - in_lo;
    int no = (out + 63) >> 6;
    int ni = (span + 63) >> 6;
    int total_ch = ni * no;
    float* ws0 = lds + PW_0;
    float* ws1 = lds + PW_1;
    float wreg[16];
    auto preload = [&](int t) {
        int i0 = (t / no) << 6, oc = (t % no) << 6;
        int gi = in_lo + i0 + ii16;
#pragma unroll
        for (int u = 0; u < 16; ++u) {
            int go = oc + oo16 + 4 * u;
            wreg[u] = (gi < in_hi && go < out)
                ? wt[(long)gi * out + go] : 0.f;
        }
    };
    preload(0);
    float acc = 0.f;
    const float* zr = dz_l + rq * PXMAX;
    for (int t = 0; t < total_ch; ++t) {
        int ii = t / no, oi = t % no;
        float* ws = (t & 1) ? ws1 : ws0;
        float wb[16];
#pragma unroll
        for (int u = 0; u < 16; ++u) wb[u] = wreg[u];
        if (t + 1 < total_ch) preload(t + 1);
#pragma unroll
        for (int u = 0; u < 16; ++u)
            ws[(oo16 + 4 * u) * 65 + ii16] = wb[u];
        __syncthreads();
        if (oi == 0) acc = 0.f;
        int oc = oi << 6;
#pragma unroll 8
        for (int o = 0; o < 64; ++o)
            acc += zr[oc + o] * ws[o * 65 + c];
        if (oi == no - 1) {
            int i = (ii << 6) + c;
            int pad_to = ni << 6;
            bool live = (i < span) && (r0 + rq < B);
            float v = acc;
            if (live && hsave)
                v *= act_mask(prev_act, hsave[(r0 + rq) * span + i]);
            if (i < pad_to)
                dx_l[rq * PXMAX + i] = (i < span) ? v : 0.f;
            if (live && dxsave) dxsave[(r0 + rq) * span + i] = v;
        }
        __syncthreads();
    }
}

// projection + CE for PR LDS-resident p_t rows (wave rq owns row r0+rq;
// scratch m rows in m_l)
__device__ inline void p_proj_ce_rows(const PStepArgs& g, const float* pt_l,
                                      float* m_l, long r0) {
    int rq = threadIdx.x >> 6, lane = threadIdx.x & 63;
    long row = r0 + rq;
    int K = g.K;
    float* mrow = m_l + rq * PXMAX;
    mrow[lane] = 0.f;
    __builtin_amdgcn_wave_barrier();
    bool live = row < g.B;
    if (live && lane < K) {
        float delta = (g.v_max - g.v_min) / (K - 1);
        float z = g.v_min + lane * delta;
        float tz = g.br[row] + g.gamma_n * (1.f - g.bd[row]) * z;
        tz = fminf(g.v_max, fmaxf(g.v_min, tz));
        float b = (tz - g.v_min) / delta;
        int l = (int)floorf(b), u = (int)ceilf(b);
        if (l == u) { if (u > 0) l -= 1; else u += 1; }
        float p = pt_l[rq * PXMAX + lane];
        atomicAdd(&mrow[l], p * ((float)u - b));
        atomicAdd(&mrow[u], p * (b - (float)l));
    }
    __builtin_amdgcn_wave_barrier();
    if (!live) return;
    float mv = (lane < K) ? mrow[lane] : 0.f;
    float qv = (lane < K) ? g.q[row * K + lane] : 0.f;
    if (lane < K) g.m_proj[row * K + lane] = mv;
    float dot = mv * qv;
    float ce = -mv * __logf(qv + 1e-10f);
    for (int s = 32; s > 0; s >>= 1) {
        dot += __shfl_xor(dot, s, 64);
        ce += __shfl_xor(ce, s, 64);
    }
    float scale = (g.is_weighting && g.bw) ? g.bw[row] : 1.f;
    if (lane < K)
        g.dlog[row * K + lane] = scale * (qv - mv) / (float)g.B;
    if (lane == 0) {
        g.pri[row] = dot + g.per_eps;
        atomicAdd(&g.cnt->loss_critic, scale * ce / (float)g.B);
    }
}

// policy-head gradient for PR LDS-resident pq rows -> pd rows (zeroed pad)
__device__ inline void p_pgrad_rows(const PStepArgs& g, const float* pq_l,
                                    float* pd_l, long r0) {
    int rq = threadIdx.x >> 6, lane = threadIdx.x & 63;
    long row = r0 + rq;
    int K = g.K;
    bool live = row < g.B;
    float delta = (g.v_max - g.v_min) / (K - 1);
    float z = g.v_min + lane * delta;
    float qv = (live && lane < K) ? pq_l[rq * PXMAX + lane] : 0.f;
    float e = qv * z;
    for (int s = 32; s > 0; s >>= 1) e += __shfl_xor(e, s, 64);
    pd_l[rq * PXMAX + lane] = (live && lane < K)
        ? -qv * (z - e) / (float)g.B : 0.f;
    if (live && lane == 0)
        atomicAdd(&g.cnt->loss_actor, -e / (float)g.B);
}

__global__ void __launch_bounds__(256, 1)
k_step_chain(PStepArgs g, int nsteps) {
    __shared__ float lds[PLDS_FLOATS];
    unsigned long long tgt = g.gbar[160];
    unsigned long long* ctr = g.gbar;
    int wg = blockIdx.x;
    NetPtrs a = net_ptrs(g.p_actor, g.al);
    NetPtrs at = net_ptrs(g.p_actor_t, g.al);
    NetPtrs c = net_ptrs(g.p_critic, g.cl);
    NetPtrs ct = net_ptrs(g.p_critic_t, g.cl);
    const int O = g.O, A = g.A, H = g.H, K = g.K, B = g.B;
    const int ngr = (B + PR - 1) / PR;
    float* XA = lds + PX_A;
    float* XB = lds + PX_B;
    const int padO = (O + 63) & ~63;
    const int padHA = (H + A + 63) & ~63;

    for (int s = 0; s < nsteps; ++s) {
        PTIME(g, s, 32);
        p_sample(g);
        p_bar(ctr, tgt); PTIME(g, s, 33);
        // C1: three independent forward chains
        if (wg < ngr) {                        // actor_target(s2) -> a2
            long r0 = (long)wg * PR;
            p_rows_load(XA, g.bs2, r0, B, O, 0, padO);
            p_layer_rows(lds, XA, XB, at.w1, at.b1, O, H, ACT_RELU,
                         nullptr, r0, B);
            p_layer_rows(lds, XB, XA, at.w2, at.b2, H, H, ACT_NONE,
                         nullptr, r0, B);
            p_layer_rows(lds, XA, XB, at.w3, at.b3, H, H, ACT_RELU,
                         nullptr, r0, B);
            p_layer_rows(lds, XB, XA, at.w4, at.b4, H, A, ACT_TANH,
                         g.a2, r0, B);
        } else if (wg < 2 * ngr) {             // critic(s, a) -> q
            long r0 = (long)(wg - ngr) * PR;
            p_rows_load(XA, g.bs, r0, B, O, 0, padO);
            p_layer_rows(lds, XA, XB, c.w1, c.b1, O, H, ACT_RELU,
                         g.c_h1, r0, B);
            p_rows_load(XB, g.ba, r0, B, A, H, padHA);
            p_layer_rows(lds, XB, XA, c.w2, c.b2, H + A, H, ACT_RELU,
                         g.c_h2, r0, B);
            p_layer_rows(lds, XA, XB, c.w3, c.b3, H, H, ACT_RELU,
                         g.c_h3, r0, B);
            p_layer_rows(lds, XB, XA, c.w4, c.b4, H, K, ACT_SOFTMAX,
                         g.q, r0, B);
        } else if (wg < 3 * ngr) {             // actor(s) -> a_out
            long r0 = (long)(wg - 2 * ngr) * PR;
            p_rows_load(XA, g.bs, r0, B, O, 0, padO);
            p_layer_rows(lds, XA, XB, a.w1, a.b1, O, H, ACT_RELU,
                         g.pa_h1, r0, B);
            p_layer_rows(lds, XB, XA, a.w2, a.b2, H, H, ACT_NONE,
                         g.pa_h2, r0, B);
            p_layer_rows(lds, XA, XB, a.w3, a.b3, H, H, ACT_RELU,
                         g.pa_h3, r0, B);
            p_layer_rows(lds, XB, XA, a.w4, a.b4, H, A, ACT_TANH,
                         g.a_out, r0, B);
        }
        p_bar(ctr, tgt); PTIME(g, s, 34);
        // C2: critic_target(s2, a2) chain + projection + CE (row-local)
        if (wg < ngr) {
            long r0 = (long)wg * PR;
            p_rows_load(XA, g.bs2, r0, B, O, 0, padO);
            p_layer_rows(lds, XA, XB, ct.w1, ct.b1, O, H, ACT_RELU,
                         nullptr, r0, B);
            p_rows_load(XB, g.a2, r0, B, A, H, padHA);
            p_layer_rows(lds, XB, XA, ct.w2, ct.b2, H + A, H, ACT_RELU,
                         nullptr, r0, B);
            p_layer_rows(lds, XA, XB, ct.w3, ct.b3, H, H, ACT_RELU,
                         nullptr, r0, B);
            p_layer_rows(lds, XB, XA, ct.w4, ct.b4, H, K, ACT_SOFTMAX,
                         g.p_t, r0, B);
            p_proj_ce_rows(g, XA, XB, r0);
        }
        p_bar(ctr, tgt); PTIME(g, s, 35);
        // C3: critic backward-dX chain (pre-update weights)
        if (wg < ngr) {
            long r0 = (long)wg * PR;
            p_rows_load(XA, g.dlog, r0, B, K, 0, 64);
            p_dx_rows(lds, XA, XB, c.w4, 0, H, K, g.c_h3, ACT_RELU,
                      g.d3, r0, B);
            p_dx_rows(lds, XB, XA, c.w3, 0, H, H, g.c_h2, ACT_RELU,
                      g.d2, r0, B);
            p_dx_rows(lds, XA, XB, c.w2, 0, H, H, g.c_h1, ACT_RELU,
                      g.d1, r0, B);
        }
        p_bar(ctr, tgt); PTIME(g, s, 36);
        // C4: critic dW (whole grid)
        if (wg < 4)
            p_dw2(lds, g.d1, g.bs, nullptr, g.g_critic + g.cl[0].w_off,
                  g.g_critic + g.cl[0].b_off, B, O, 0, H, wg, 4);
        else if (wg < 32)
            p_dw2(lds, g.d2, g.c_h1, g.ba, g.g_critic + g.cl[1].w_off,
                  g.g_critic + g.cl[1].b_off, B, H, A, H, wg - 4, 28);
        else if (wg < 56)
            p_dw2(lds, g.d3, g.c_h2, nullptr, g.g_critic + g.cl[2].w_off,
                  g.g_critic + g.cl[2].b_off, B, H, 0, H, wg - 32, 24);
        else
            p_dw2(lds, g.dlog, g.c_h3, nullptr, g.g_critic + g.cl[3].w_off,
                  g.g_critic + g.cl[3].b_off, B, H, 0, K, wg - 56, 8);
        p_bar(ctr, tgt); PTIME(g, s, 37);
        // C5: critic Adam + target soft-update
        p_adam_lerp(g.p_critic, g.g_critic, g.m_critic, g.v_critic,
                    g.p_critic_t, g.n_critic, g.lr_critic, g.tau,
                    g.cnt->adam_t_critic);
        p_bar(ctr, tgt); PTIME(g, s, 38);
        // C6: policy megachain (UPDATED critic), with the PER tree
        // write-back overlapped on the last workgroup
        if (wg < ngr) {
            long r0 = (long)wg * PR;
            p_rows_load(XA, g.bs, r0, B, O, 0, padO);
            p_layer_rows(lds, XA, XB, c.w1, c.b1, O, H, ACT_RELU,
                         g.pc_h1, r0, B);
            p_rows_load(XB, g.a_out, r0, B, A, H, padHA);
            p_layer_rows(lds, XB, XA, c.w2, c.b2, H + A, H, ACT_RELU,
                         g.pc_h2, r0, B);
            p_layer_rows(lds, XA, XB, c.w3, c.b3, H, H, ACT_RELU,
                         g.pc_h3, r0, B);
            p_layer_rows(lds, XB, XA, c.w4, c.b4, H, K, ACT_SOFTMAX,
                         g.pq, r0, B);
            p_pgrad_rows(g, XA, XB, r0);
            p_dx_rows(lds, XB, XA, c.w4, 0, H, K, g.pc_h3, ACT_RELU,
                      nullptr, r0, B);
            p_dx_rows(lds, XA, XB, c.w3, 0, H, H, g.pc_h2, ACT_RELU,
                      nullptr, r0, B);
            p_dx_rows(lds, XB, XA, c.w2, H, H + A, H, g.a_out, ACT_TANH,
                      g.adz, r0, B);
            p_dx_rows(lds, XA, XB, a.w4, 0, H, A, g.pa_h3, ACT_RELU,
                      g.az3, r0, B);
            p_dx_rows(lds, XB, XA, a.w3, 0, H, H, nullptr, ACT_NONE,
                      g.az2, r0, B);
            p_dx_rows(lds, XA, XB, a.w2, 0, H, H, g.pa_h1, ACT_RELU,
                      g.az1, r0, B);
        } else if (wg == PNWG - 1) {
            p_per_update(g, false, PNWG - 1);
        }
        p_bar(ctr, tgt); PTIME(g, s, 39);
        // C7: actor dW
        if (wg < 4)
            p_dw2(lds, g.az1, g.bs, nullptr, g.g_actor + g.al[0].w_off,
                  g.g_actor + g.al[0].b_off, B, O, 0, H, wg, 4);
        else if (wg < 30)
            p_dw2(lds, g.az2, g.pa_h1, nullptr, g.g_actor + g.al[1].w_off,
                  g.g_actor + g.al[1].b_off, B, H, 0, H, wg - 4, 26);
        else if (wg < 56)
            p_dw2(lds, g.az3, g.pa_h2, nullptr, g.g_actor + g.al[2].w_off,
                  g.g_actor + g.al[2].b_off, B, H, 0, H, wg - 30, 26);
        else
            p_dw2(lds, g.adz, g.pa_h3, nullptr, g.g_actor + g.al[3].w_off,
                  g.g_actor + g.al[3].b_off, B, H, 0, A, wg - 56, 8);
        p_bar(ctr, tgt); PTIME(g, s, 40);
        // C8: actor Adam + soft-update
        p_adam_lerp(g.p_actor, g.g_actor, g.m_actor, g.v_actor,
                    g.p_actor_t, g.n_actor, g.lr_actor, g.tau,
                    g.cnt->adam_t_actor);
        p_bar(ctr, tgt); PTIME(g, s, 41);
        // C9: schedule-counter tick (after both Adams read their t)
        if (wg == 0 && threadIdx.x == 0) {
            g.cnt->beta_t += 1;
            g.cnt->adam_t_actor += 1;
            g.cnt->adam_t_critic += 1;
            g.cnt->rng_epoch += 1;
        }
        p_bar(ctr, tgt); PTIME(g, s, 42);
    }
    if (wg == 0 && threadIdx.x == 0) g.gbar[160] = tgt;
}

__global__ void __launch_bounds__(256, 1)
k_step_persistent(PStepArgs g, int nsteps) {
    __shared__ float lds[PLDS_FLOATS];
    unsigned long long tgt = g.gbar[160];   // cross-launch round base
    unsigned long long* ctr = g.gbar;
    int wg = blockIdx.x;
    // quad-barrier round base: my quad's arrival counter as of launch
    // entry (every member reads it before ANY member can arrive — the
    // first q_bar use sits behind several grid barriers)
    unsigned long long qrnd = ctr[192 + ((wg >> 2) << 4)];
    NetPtrs a = net_ptrs(g.p_actor, g.al);
    NetPtrs at = net_ptrs(g.p_actor_t, g.al);
    NetPtrs c = net_ptrs(g.p_critic, g.cl);
    NetPtrs ct = net_ptrs(g.p_critic_t, g.cl);
    const int O = g.O, A = g.A, H = g.H, K = g.K, B = g.B;

    // Schedule counters are read ONCE per launch and threaded through as
    // base+s (saves the per-step tick phase + its grid barrier, ~7 us);
    // wg 0 writes the advanced values back after the last step.
    const long long bt0 = g.cnt->beta_t;
    const long long ep0 = g.cnt->rng_epoch;
    const long long ta0 = g.cnt->adam_t_actor;
    const long long tc0 = g.cnt->adam_t_critic;
    // bs is double-buffered by step parity so the policy-block tail crew
    // can pre-sample the NEXT step's batch while this step's bs is still
    // read by the policy forward and the actor dW.  Parity derives from
    // the beta counter, so resume/relaunch stay deterministic.
    const int QW = ((B + 3) / 4) * 4;        // wgs used by the quad chains
    const bool otail = (QW + 32 <= PNWG) && (B <= 128);
    unsigned long long trnd = ctr[1280];

    for (int s = 0; s < nsteps; ++s) {
        float* bs_cur = ((bt0 + s) & 1) ? g.bs_b : g.bs;
        float* bs_nxt = ((bt0 + s) & 1) ? g.bs : g.bs_b;
               PTIME(g, s, 0);
        // PH0: PER sample + batch gather (only on a launch's first step —
        // later steps were pre-sampled under the previous actor-Adam,
        // whose 64 overlap wgs cover 256 probes)
        if (s == 0 || B > (PNWG - PNWG * 3 / 4) * 4) {
            p_sample(g, 0, 0, ep0 + s, bt0 + s, bs_cur);
            p_bar(ctr, tgt);
        }
        PTIME(g, s, 1);
        // PH1: four independent L1s (16 wgs each)
        if (wg < PW4)
            p_fwd(lds, g.bs2, nullptr, at.w1, at.b1, g.at_h1, B, O, 0, H,
                  ACT_RELU, wg, PW4);
        else if (wg < 2 * PW4) {
            if (!otail || s == 0)     // else precomputed by the tail crew
                p_fwd(lds, g.bs2, nullptr, ct.w1, ct.b1, g.ct_h1, B, O, 0,
                      H, ACT_RELU, wg - PW4, PW4);
        } else if (wg < 3 * PW4) {
            if (!otail || s == 0)
                p_fwd(lds, bs_cur, nullptr, c.w1, c.b1, g.c_h1, B, O, 0, H,
                      ACT_RELU, wg - 2 * PW4, PW4);
        } else
            p_fwd(lds, bs_cur, nullptr, a.w1, a.b1, g.pa_h1, B, O, 0, H,
                  ACT_RELU, wg - 3 * PW4, PW4);
        p_bar(ctr, tgt); PTIME(g, s, 2);
        // PH2: actor_t.L2 | critic.L2(cat h1, a) | actor.L2
        if (wg < PW3A)
            p_fwd(lds, g.at_h1, nullptr, at.w2, at.b2, g.at_h2, B, H, 0, H,
                  ACT_NONE, wg, PW3A);
        else if (wg < 2 * PW3A)
            p_fwd(lds, g.c_h1, g.ba, c.w2, c.b2, g.c_h2, B, H, A, H,
                  ACT_RELU, wg - PW3A, PW3A);
        else
            p_fwd(lds, g.pa_h1, nullptr, a.w2, a.b2, g.pa_h2, B, H, 0, H,
                  ACT_NONE, wg - 2 * PW3A, PW3B);
        p_bar(ctr, tgt); PTIME(g, s, 3);
        // PH3: L3s
        if (wg < PW3A)
            p_fwd(lds, g.at_h2, nullptr, at.w3, at.b3, g.at_h3, B, H, 0, H,
                  ACT_RELU, wg, PW3A);
        else if (wg < 2 * PW3A)
            p_fwd(lds, g.c_h2, nullptr, c.w3, c.b3, g.c_h3, B, H, 0, H,
                  ACT_RELU, wg - PW3A, PW3A);
        else
            p_fwd(lds, g.pa_h2, nullptr, a.w3, a.b3, g.pa_h3, B, H, 0, H,
                  ACT_RELU, wg - 2 * PW3A, PW3B);
        p_bar(ctr, tgt); PTIME(g, s, 4);
        // PH4: heads — actor_t tanh -> a2 | critic softmax -> q |
        //      actor tanh -> a_out
        if (wg < PW3A)
            p_fwd(lds, g.at_h3, nullptr, at.w4, at.b4, g.a2, B, H, 0, A,
                  ACT_TANH, wg, PW3A);
        else if (wg < 2 * PW3A)
            p_fwd(lds, g.c_h3, nullptr, c.w4, c.b4, g.q, B, H, 0, K,
                  ACT_SOFTMAX, wg - PW3A, PW3A);
        else
            p_fwd(lds, g.pa_h3, nullptr, a.w4, a.b4, g.a_out, B, H, 0, A,
                  ACT_TANH, wg - 2 * PW3A, PW3B);
        // per-step loss accumulators zeroed here (grid barrier below
        // orders this before the quad-chained proj-CE / pgrad adds)
        if (wg == PNWG - 1 && threadIdx.x == 0) {
            g.cnt->loss_critic = 0.f;
            g.cnt->loss_actor = 0.f;
        }
        p_bar(ctr, tgt); PTIME(g, s, 5);
        // PH5-8 as ONE phase: the critic_target chain + projection + CE is
        // ROW-LOCAL (row b's p_t depends only on row b's ct_h1/a2), so each
        // quad of 4 wgs walks its 4-row group through ct.L2 -> ct.L3 ->
        // ct.L4 softmax -> C51 projection/CE with quad-local barriers; the
        // 16 quads never wait on each other (grid barriers cost ~3.3 us,
        // quad barriers well under 1 us, and straggler coupling is gone).
        {
            int q = wg >> 2, m = wg & 3;
            unsigned long long* qc = ctr + 192 + (q << 4);
            for (int rg = q; rg < (B + 3) / 4; rg += PNWG / 4) {
                long r0 = (long)rg * 4;
                int nb = min(4, B - (int)r0);
                p_fwd(lds, g.ct_h1 + r0 * H, g.a2 + r0 * A, ct.w2, ct.b2,
                      g.ct_h2 + r0 * H, nb, H, A, H, ACT_RELU, m, 4);
                q_bar(qc, qrnd);
                p_fwd(lds, g.ct_h2 + r0 * H, nullptr, ct.w3, ct.b3,
                      g.ct_h3 + r0 * H, nb, H, 0, H, ACT_RELU, m, 4);
                q_bar(qc, qrnd);
                p_fwd(lds, g.ct_h3 + r0 * H, nullptr, ct.w4, ct.b4,
                      g.p_t + r0 * K, nb, H, 0, K, ACT_SOFTMAX, m, 4);
                if (m == 0)
                    p_project_ce(g, lds, (int)r0);
                // no trailing q_bar: row groups are independent and the
                // closing grid barrier publishes everything
            }
        }
        p_bar(ctr, tgt);
        PTIME(g, s, 6); PTIME(g, s, 7); PTIME(g, s, 8); PTIME(g, s, 9);
        // PH10-12 as ONE phase: critic dX chain (pre-update weights),
        // row-local — quad-chained like the target block above
        {
            int q = wg >> 2, m = wg & 3;
            unsigned long long* qc = ctr + 192 + (q << 4);
            for (int rg = q; rg < (B + 3) / 4; rg += PNWG / 4) {
                long r0 = (long)rg * 4;
                int nb = min(4, B - (int)r0);
                p_bwd_dx(lds, g.dlog + r0 * K, c.w4, 0, H, K, nb,
                         g.c_h3 + r0 * H, ACT_RELU, g.d3 + r0 * H, m, 4);
                q_bar(qc, qrnd);
                p_bwd_dx(lds, g.d3 + r0 * H, c.w3, 0, H, H, nb,
                         g.c_h2 + r0 * H, ACT_RELU, g.d2 + r0 * H, m, 4);
                q_bar(qc, qrnd);
                p_bwd_dx(lds, g.d2 + r0 * H, c.w2, 0, H, H, nb,
                         g.c_h1 + r0 * H, ACT_RELU, g.d1 + r0 * H, m, 4);
            }
        }
        p_bar(ctr, tgt);
        PTIME(g, s, 10); PTIME(g, s, 11); PTIME(g, s, 12);
        // PH13: critic dW, 4 jobs split by tile count (l1 is the biggest)
        if (wg < PNWG / 16)
            p_dw2(lds, g.d1, bs_cur, nullptr, g.g_critic + g.cl[0].w_off,
                 g.g_critic + g.cl[0].b_off, B, O, 0, H, wg, PNWG / 16);
        else if (wg < PNWG / 2)
            p_dw2(lds, g.d2, g.c_h1, g.ba, g.g_critic + g.cl[1].w_off,
                 g.g_critic + g.cl[1].b_off, B, H, A, H, wg - PNWG / 16,
                 PNWG / 2 - PNWG / 16);
        else if (wg < PNWG * 7 / 8)
            p_dw2(lds, g.d3, g.c_h2, nullptr, g.g_critic + g.cl[2].w_off,
                 g.g_critic + g.cl[2].b_off, B, H, 0, H, wg - PNWG / 2,
                 PNWG * 7 / 8 - PNWG / 2);
        else
            p_dw2(lds, g.dlog, g.c_h3, nullptr, g.g_critic + g.cl[3].w_off,
                 g.g_critic + g.cl[3].b_off, B, H, 0, K, wg - PNWG * 7 / 8,
                 PNWG - PNWG * 7 / 8);
        p_bar(ctr, tgt); PTIME(g, s, 13);
        // PH14: Adam + target soft-update, critic
        p_adam_lerp(g.p_critic, g.g_critic, g.m_critic, g.v_critic,
                    g.p_critic_t, g.n_critic, g.lr_critic, g.tau,
                    tc0 + s);
        p_bar(ctr, tgt); PTIME(g, s, 14);
        // PH15-25 as ONE phase: the whole policy megachain — critic(s,
        // a_out) forward with UPDATED critic params, policy head gradient,
        // dX back through the critic to the action slice, then dX through
        // the actor — is row-local end-to-end, so each quad chains its
        // 4-row group through all 10 layers with quad-local barriers
        // (replaces 11 grid barriers; VERDICT r1 #7 / NOTES.md roadmap —
        // this is the pair-barrier idea at quad granularity, which keeps
        // the weight re-streaming of the tiled form: one full W read per
        // row-group, L2-resident)
        {
            int q = wg >> 2, m = wg & 3;
            unsigned long long* qc = ctr + 192 + (q << 4);
            if (otail && wg >= QW && wg < QW + 32) {
                // tail crew (32 spare wgs): this step's PER write-back,
                // then the NEXT step's sample and its ct/c L1 forwards —
                // all hidden under the ~70 us policy megachain.  Safe:
                // bs2/ba/br/bd/bw/bidx of step s were fully consumed
                // before this block; bs_nxt is the other bs buffer.
                p_per_update(g, false, QW);
                t_bar(ctr + 1280, trnd, 32);
                if (s + 1 < nsteps) {
                    p_sample(g, QW, 1, ep0 + s + 1, bt0 + s + 1, bs_nxt);
                    t_bar(ctr + 1280, trnd, 32);
                    p_fwd(lds, g.bs2, nullptr, ct.w1, ct.b1, g.ct_h1, B,
                          O, 0, H, ACT_RELU, wg - QW, 32);
                    p_fwd(lds, bs_nxt, nullptr, c.w1, c.b1, g.c_h1, B, O,
                          0, H, ACT_RELU, wg - QW, 32);
                }
            }
            for (int rg = q; rg < (B + 3) / 4; rg += PNWG / 4) {
                long r0 = (long)rg * 4;
                int nb = min(4, B - (int)r0);
                PTIME(g, s, 40);
                p_fwd(lds, bs_cur + r0 * O, nullptr, c.w1, c.b1,
                      g.pc_h1 + r0 * H, nb, O, 0, H, ACT_RELU, m, 4);
                q_bar(qc, qrnd); PTIME(g, s, 41);
                p_fwd(lds, g.pc_h1 + r0 * H, g.a_out + r0 * A, c.w2, c.b2,
                      g.pc_h2 + r0 * H, nb, H, A, H, ACT_RELU, m, 4);
                q_bar(qc, qrnd); PTIME(g, s, 42);
                p_fwd(lds, g.pc_h2 + r0 * H, nullptr, c.w3, c.b3,
                      g.pc_h3 + r0 * H, nb, H, 0, H, ACT_RELU, m, 4);
                q_bar(qc, qrnd); PTIME(g, s, 43);
                p_fwd(lds, g.pc_h3 + r0 * H, nullptr, c.w4, c.b4,
                      g.pq + r0 * K, nb, H, 0, K, ACT_SOFTMAX, m, 4);
                if (m == 0)
                    p_policy_grad(g, (int)r0);
                q_bar(qc, qrnd); PTIME(g, s, 44);
                p_bwd_dx(lds, g.pd3 + r0 * K, c.w4, 0, H, K, nb,
                         g.pc_h3 + r0 * H, ACT_RELU, g.pd2 + r0 * H, m, 4);
                q_bar(qc, qrnd); PTIME(g, s, 45);
                p_bwd_dx(lds, g.pd2 + r0 * H, c.w3, 0, H, H, nb,
                         g.pc_h2 + r0 * H, ACT_RELU, g.pdh1 + r0 * H,
                         m, 4);
                q_bar(qc, qrnd); PTIME(g, s, 46);
                if (m == 0)
                    p_bwd_dx_narrow(g, g.pdh1, c.w2, H, H + A, H,
                                    g.a_out, g.adz, (int)r0);
                q_bar(qc, qrnd); PTIME(g, s, 47);
                p_bwd_dx(lds, g.adz + r0 * A, a.w4, 0, H, A, nb,
                         g.pa_h3 + r0 * H, ACT_RELU, g.az3 + r0 * H, m, 4);
                q_bar(qc, qrnd); PTIME(g, s, 48);
                p_bwd_dx(lds, g.az3 + r0 * H, a.w3, 0, H, H, nb,
                         nullptr, ACT_NONE, g.az2 + r0 * H, m, 4);
                q_bar(qc, qrnd); PTIME(g, s, 49);
                p_bwd_dx(lds, g.az2 + r0 * H, a.w2, 0, H, H, nb,
                         g.pa_h1 + r0 * H, ACT_RELU, g.az1 + r0 * H, m, 4);
                PTIME(g, s, 51);
            }
        }
        p_bar(ctr, tgt);
        PTIME(g, s, 15); PTIME(g, s, 16); PTIME(g, s, 17);
        PTIME(g, s, 18); PTIME(g, s, 19); PTIME(g, s, 20);
        PTIME(g, s, 21); PTIME(g, s, 22); PTIME(g, s, 23);
        PTIME(g, s, 24); PTIME(g, s, 25);
        // PH26: actor dW, with the PER tree write-back overlapped on the
        // last workgroup (priorities have been final since the proj+CE
        // phase; the counter tick moves to the final phase so both Adams
        // still read this step's t)
        if (wg < PNWG / 16)
            p_dw2(lds, g.az1, bs_cur, nullptr, g.g_actor + g.al[0].w_off,
                 g.g_actor + g.al[0].b_off, B, O, 0, H, wg, PNWG / 16);
        else if (wg < PNWG * 15 / 32)
            p_dw2(lds, g.az2, g.pa_h1, nullptr, g.g_actor + g.al[1].w_off,
                 g.g_actor + g.al[1].b_off, B, H, 0, H, wg - PNWG / 16,
                 PNWG * 15 / 32 - PNWG / 16);
        else if (wg < PNWG * 7 / 8)
            p_dw2(lds, g.az3, g.pa_h2, nullptr, g.g_actor + g.al[2].w_off,
                 g.g_actor + g.al[2].b_off, B, H, 0, H, wg - PNWG * 15 / 32,
                 PNWG * 7 / 8 - PNWG * 15 / 32);
        else if (wg < PNWG - 2)
            p_dw2(lds, g.adz, g.pa_h3, nullptr, g.g_actor + g.al[3].w_off,
                 g.g_actor + g.al[3].b_off, B, H, 0, A, wg - PNWG * 7 / 8,
                 PNWG - 2 - PNWG * 7 / 8);
        else if (!otail && wg == PNWG - 1)
            p_per_update(g, false, PNWG - 1);   // tail crew did it when otail
        p_bar(ctr, tgt); PTIME(g, s, 26);
        // PH27: Adam + soft-update, actor (wgs 0-47) overlapped with the
        // NEXT step's PER sample + gather (wgs 48-63; tree was repaired in
        // PH26, schedule counters offset by the pending tick)
        {
            int anw = otail ? PNWG : PNWG * 3 / 4;
            if (wg < anw)
                p_adam_lerp(g.p_actor, g.g_actor, g.m_actor, g.v_actor,
                            g.p_actor_t, g.n_actor, g.lr_actor, g.tau,
                            ta0 + s, anw);
            else if (s + 1 < nsteps && B <= (PNWG - PNWG * 3 / 4) * 4)
                p_sample(g, PNWG * 3 / 4, 1, ep0 + s + 1, bt0 + s + 1,
                         bs_nxt);
        }
        p_bar(ctr, tgt); PTIME(g, s, 27); PTIME(g, s, 28);
        // (the per-step schedule-counter tick phase is gone: counters were
        // hoisted to base+s registers; final values land below)
    }
    if (wg == 0 && threadIdx.x == 0) {
        g.cnt->beta_t = bt0 + nsteps;
        g.cnt->rng_epoch = ep0 + nsteps;
        g.cnt->adam_t_actor = ta0 + nsteps;
        g.cnt->adam_t_critic = tc0 + nsteps;
        g.gbar[160] = tgt;
    }
}

// ===========================================================================
// Host engine
// ===========================================================================

static long next_pow2(long n) { long c = 1; while (c < n) c <<= 1; return c; }
static int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

struct Net {
    LayerDesc l[4];
    long n_params;
};

static Net make_actor_net(int O, int A, int H) {
    Net n{};
    long off = 0;
    auto add = [&](int i1, int i2, int o) {
        LayerDesc d{i1, i2, o, off, off + (long)(i1 + i2) * o};
        off += (long)(i1 + i2) * o + o;
        return d;
    };
    n.l[0] = add(O, 0, H);
    n.l[1] = add(H, 0, H);
    n.l[2] = add(H, 0, H);
    n.l[3] = add(H, 0, A);
    n.n_params = off;
    return n;
}

static Net make_critic_net(int O, int A, int H, int K) {
    Net n{};
    long off = 0;
    auto add = [&](int i1, int i2, int o) {
        LayerDesc d{i1, i2, o, off, off + (long)(i1 + i2) * o};
        off += (long)(i1 + i2) * o + o;
        return d;
    };
    n.l[0] = add(O, 0, H);
    n.l[1] = add(H, A, H);
    n.l[2] = add(H, 0, H);
    n.l[3] = add(H, 0, K);
    n.n_params = off;
    return n;
}

class Engine {
public:
    EngineCfg cfg;
    Net anet, cnet;
    long tree_cap;
    hipStream_t stream;
    hipGraph_t graph = nullptr;
    hipGraphExec_t graph_exec = nullptr;
    int graph_steps = 0;

    // device buffers
    float *p_actor, *p_actor_t, *p_critic, *p_critic_t;
    float *g_actor, *g_critic;
    float *m_actor, *v_actor, *m_critic, *v_critic;
    float *rs, *ra, *rr, *rs2, *rd;                 // replay SoA
    double *sum_tree, *min_tree;
    Counters* cnt;
    // batch + activations workspace
    float *bs, *bs_b, *ba, *br, *bs2, *bd, *bw, *pri;
    long *bidx;
    float *at_h1, *at_h2, *at_h3, *a2;              // actor_target path
    float *ct_h1, *ct_h2, *ct_h3, *p_t, *m_proj;    // critic_target path
    float *c_h1, *c_h2, *c_h3, *q;                  // critic path
    float *dlog, *d3, *d2, *d1, *da;                // critic backward deltas
    float *logits;                                  // wide softmax scratch
    float *part_scratch;                            // split-K partial sums
    float *pa_h1, *pa_h2, *pa_h3, *a_out;           // actor (policy) path
    float *pc_h1, *pc_h2, *pc_h3, *pq;              // critic(s, actor(s))
    float *pd3, *pd2, *pdh1, *pda, *adz;            // policy backward deltas
    float *az1, *az2, *az3;                         // actor per-layer dz rows
    float *ing_s, *ing_a, *ing_r, *ing_s2, *ing_d;  // ingestion staging
    float* dw_parts;                                // split-K dW partials
    int ing_cap;
    unsigned long long* gbar;                       // persistent grid barrier
    unsigned long long* tstamp;                     // phase timing probe

    Engine(const EngineCfg& c) : cfg(c) {
        if (c.hidden + c.act > FWD_XMAX || c.obs > FWD_XMAX)
            throw std::runtime_error("layer fan-in exceeds FWD_XMAX LDS tile");
        if (c.atoms > 64)
            throw std::runtime_error("n_atoms > 64 (softmax is wave-wide)");
        anet = make_actor_net(c.obs, c.act, c.hidden);
        cnet = make_critic_net(c.obs, c.act, c.hidden, c.atoms);
        tree_cap = next_pow2(c.capacity);
        HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
        // co-residency gate for the persistent path: all PNWG workgroups
        // must fit on the device at once or the software grid barrier
        // cannot make progress
        int dev = 0, per_cu = 0;
        HIP_CHECK(hipGetDevice(&dev));
        hipDeviceProp_t prop;
        HIP_CHECK(hipGetDeviceProperties(&prop, dev));
        HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
            &per_cu, k_step_persistent, 256, 0));
        persistent_fits_ = (long)per_cu * prop.multiProcessorCount >= PNWG;
        HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
            &per_cu, k_step_chain, 256, 0));
        chain_fits_ = (long)per_cu * prop.multiProcessorCount >= PNWG;
        device_ = dev;
        alloc();
    }

    int device_ = 0;

    bool persistent_fits_ = false;
    bool chain_fits_ = false;

    // the persistent grid barrier's timeout flag (gbar[176]); raised on
    // device when a p_bar spins out, fatal host-side — the engine state
    // must be considered corrupt after it
    void check_bar_error() {
        unsigned long long e = 0;
        HIP_CHECK(hipMemcpy(&e, gbar + 176, 8, hipMemcpyDeviceToHost));
        if (e)
            throw std::runtime_error(
                "persistent grid barrier timed out (desynchronized "
                "workgroups; device state is corrupt) — occupancy check "
                "passed but the grid did not make progress");
    }

    ~Engine() {
        rollout_free();
        if (graph_exec) hipGraphExecDestroy(graph_exec);
        if (graph) hipGraphDestroy(graph);
        hipStreamDestroy(stream);
        hipFree(pool_);
    }

    void* pool_ = nullptr;

    template <typename T>
    T* carve(long nelem, long& off) {
        long bytes = nelem * (long)sizeof(T);
        bytes = (bytes + 255) & ~255L;
        T* p = reinterpret_cast<T*>(static_cast<char*>(pool_) + off);
        off += bytes;
        return p;
    }

    // One carve sequence, executed twice (count pass with null base, then
    // assign pass) so sizes and pointers can never drift apart.
    long layout() {
        const int B = cfg.batch, O = cfg.obs, A = cfg.act, H = cfg.hidden,
                  K = cfg.atoms;
        const long C = cfg.capacity;
        const long pa = anet.n_params, pc = cnet.n_params;
        ing_cap = 65536;
        long off = 0;
        p_actor = carve<float>(pa, off);  p_actor_t = carve<float>(pa, off);
        g_actor = carve<float>(pa, off);
        p_critic = carve<float>(pc, off); p_critic_t = carve<float>(pc, off);
        g_critic = carve<float>(pc, off);
        m_actor = carve<float>(pa, off);  v_actor = carve<float>(pa, off);
        m_critic = carve<float>(pc, off); v_critic = carve<float>(pc, off);
        rs = carve<float>(C * O, off); ra = carve<float>(C * A, off);
        rr = carve<float>(C, off); rs2 = carve<float>(C * O, off);
        rd = carve<float>(C, off);
        sum_tree = carve<double>(2 * tree_cap, off);
        min_tree = carve<double>(2 * tree_cap, off);
        cnt = carve<Counters>(1, off);
        bs = carve<float>((long)B * O, off);
        bs_b = carve<float>((long)B * O, off);
        ba = carve<float>((long)B * A, off);
        br = carve<float>(B, off); bs2 = carve<float>((long)B * O, off);
        bd = carve<float>(B, off); bw = carve<float>(B, off);
        pri = carve<float>(B, off);
        bidx = carve<long>(B, off);
        at_h1 = carve<float>((long)B * H, off);
        at_h2 = carve<float>((long)B * H, off);
        at_h3 = carve<float>((long)B * H, off);
        a2 = carve<float>((long)B * A, off);
        ct_h1 = carve<float>((long)B * H, off);
        ct_h2 = carve<float>((long)B * H, off);
        ct_h3 = carve<float>((long)B * H, off);
        p_t = carve<float>((long)B * K, off);
        m_proj = carve<float>((long)B * K, off);
        c_h1 = carve<float>((long)B * H, off);
        c_h2 = carve<float>((long)B * H, off);
        c_h3 = carve<float>((long)B * H, off);
        q = carve<float>((long)B * K, off);
        dlog = carve<float>((long)B * K, off);
        logits = carve<float>((long)B * K, off);
        part_scratch = carve<float>((long)B * (H + A), off);
        d3 = carve<float>((long)B * H, off);
        d2 = carve<float>((long)B * H, off);
        d1 = carve<float>((long)B * H, off);
        da = carve<float>((long)B * A, off);
        pa_h1 = carve<float>((long)B * H, off);
        pa_h2 = carve<float>((long)B * H, off);
        pa_h3 = carve<float>((long)B * H, off);
        a_out = carve<float>((long)B * A, off);
        pc_h1 = carve<float>((long)B * H, off);
        pc_h2 = carve<float>((long)B * H, off);
        pc_h3 = carve<float>((long)B * H, off);
        pq = carve<float>((long)B * K, off);
        pd3 = carve<float>((long)B * K, off);
        pd2 = carve<float>((long)B * H, off);
        pdh1 = carve<float>((long)B * H, off);
        pda = carve<float>((long)B * A, off);
        adz = carve<float>((long)B * A, off);
        az1 = carve<float>((long)B * H, off);
        az2 = carve<float>((long)B * H, off);
        az3 = carve<float>((long)B * H, off);
        // split-K dW disjoint partials (wide path only): max over layers
        // of ksplit * (weights + bias); the ksplit formula mirrors
        // launch_bwd's and guarantees every segment is non-empty
        long dwp = 0;
        if (B >= 512) {
            auto dw_need = [&](int in_n, int out) {
                int ntm = ceil_div(in_n, MT_M), ntn = ceil_div(out, MT_N);
                int ks = 1;
                while (ntm * ntn * ks < 256 && ks * 2 * MT_K <= B) ks *= 2;
                long v = (long)ks * ((long)in_n * out + out);
                if (v > dwp) dwp = v;
            };
            auto need = [&](const LayerDesc& l) {
                // dW runs per input block (x1 / x2 separately)
                dw_need(l.in1, l.out);
                if (l.in2 > 0) dw_need(l.in2, l.out);
                if (l.out <= 64) {      // split-K head fwd partials
                    int it = l.in1 + l.in2;
                    int ntmh = ceil_div(B, MT_M), ksh = 1;
                    while (ntmh * ksh < 512 && ksh * 2 * MT_K <= it)
                        ksh *= 2;
                    long v2 = (long)ksh * (long)B * l.out;
                    if (v2 > dwp) dwp = v2;
                }
            };
            for (int i = 0; i < 4; ++i) { need(anet.l[i]); need(cnet.l[i]); }
        }
        dw_parts = carve<float>(dwp, off);
        gbar = carve<unsigned long long>(1536, off);
        tstamp = carve<unsigned long long>(64, off);
        ing_s = carve<float>((long)ing_cap * O, off);
        ing_a = carve<float>((long)ing_cap * A, off);
        ing_r = carve<float>(ing_cap, off);
        ing_s2 = carve<float>((long)ing_cap * O, off);
        ing_d = carve<float>(ing_cap, off);
        return off;
    }

    void alloc() {
        pool_ = nullptr;
        long total = layout();               // count pass (null base)
        HIP_CHECK(hipMalloc(&pool_, total));
        HIP_CHECK(hipMemset(pool_, 0, total));
        layout();                            // assign pass
        Counters h{};
        h.max_priority = 1.0f;
        // schedule counters pre-advanced: k_per_update increments at step
        // END for the next step, so step 1 must already read Adam t=1.
        // beta_t starts at 0 (reference LinearSchedule returns value_at(0)
        // on its first stateful call, prioritized_replay_memory.py:25-29).
        h.beta_t = 0;
        h.adam_t_actor = h.adam_t_critic = h.rng_epoch = 1;
        HIP_CHECK(hipMemcpy(cnt, &h, sizeof(Counters), hipMemcpyHostToDevice));
        // min-tree neutral element is +inf (memset zero would make every
        // unoccupied leaf the minimum and zero out all IS weights)
        hipLaunchKernelGGL(k_fill_f64, dim3(1024), dim3(256), 0, nullptr,
                           min_tree, 2 * tree_cap, HUGE_VAL);
        HIP_CHECK(hipDeviceSynchronize());
    }

    // ---------------- job builders ----------------
    FwdJob fwd_job(const float* x1, const float* x2, const float* slab,
                   const LayerDesc& l, float* y, int act_kind, int& wg) {
        FwdJob j{};
        j.x1 = x1; j.x2 = x2;
        j.wt = slab + l.w_off; j.bias = slab + l.b_off;
        j.y = y;
        j.B = cfg.batch; j.in1 = l.in1; j.in2 = l.in2; j.out = l.out;
        j.act = act_kind;
        j.wg0 = wg;
        j.nwg_b = ceil_div(cfg.batch, TB);
        j.nwg_o = ceil_div(l.out, TO);
        wg += j.nwg_b * j.nwg_o;
        return j;
    }

    bool mfma_eligible(int in_total, int out, int act_kind) const {
        // wide-batch GEMMs go to the matrix cores; narrow heads
        // (out <= 64, incl. the softmax ones) take the split-K
        // k_mfma_fwd + k_head_finish pair — their M-tile-only grids
        // were chip-starved on both k_fwd3 (~60 us) and plain MFMA
        (void)out; (void)in_total; (void)act_kind;
        return cfg.batch >= 512;
    }

    void launch_fwd(std::initializer_list<FwdJob> jobs) {
        FwdJob a[3] = {};
        int n = 0, wgs = 0;
        bool all_mfma = cfg.batch >= 512;
        for (auto& j : jobs)
            all_mfma = all_mfma && mfma_eligible(j.in1 + j.in2, j.out, j.act);
        if (all_mfma) {
            for (auto& j : jobs) {
                int in_total_j = j.in1 + j.in2;
                if (j.out <= 64) {
                    // narrow head: split-K partials + wave-per-row finish
                    int ntm = ceil_div(j.B, MT_M);
                    int ks = 1;
                    while (ntm * ks < 512 && ks * 2 * MT_K <= in_total_j)
                        ks *= 2;
                    hipLaunchKernelGGL(k_mfma_fwd, dim3(ntm * ks),
                                       dim3(256), 0, stream, j.x1, j.x2,
                                       j.wt, j.bias, j.y, j.B, j.in1,
                                       j.in2, j.out, j.act, ks, dw_parts);
                    if (ks > 1)
                        hipLaunchKernelGGL(k_head_finish, dim3(256),
                                           dim3(256), 0, stream, dw_parts,
                                           j.bias, j.y, j.B, j.out, ks,
                                           j.act);
                    continue;
                }
                // OPT-IN (D4PG_FWD2=1): measured 7% SLOWER than the
                // 64x128 tiles (272 vs 294 steps/s same-box) — the weight
                // operand is L2-resident so the halved "HBM traffic" was
                // never paid, while the split-K memset+atomics+epilogue
                // cost is real.  Kept for documentation/experimentation.
                static const bool fwd2 = [] {
                    const char* e = getenv("D4PG_FWD2");
                    return e && e[0] == '1';
                }();
                int in_total = j.in1 + j.in2;
                if (fwd2 && j.B >= 2048 && j.out >= 256 &&
                    in_total >= 256) {
                    // big GEMM: 128x256 tiles + split-K (half the weight
                    // re-streaming of the 64x128 tiles; see NOTES.md)
                    int ntm = ceil_div(j.B, M2_M);
                    int ntn = ceil_div(j.out, M2_N);
                    int ksplit = 1;
                    while (ntm * ntn * ksplit < 256 &&
                           ksplit * 2 * MT_K <= in_total)
                        ksplit *= 2;
                    HIP_CHECK(hipMemsetAsync(part_scratch, 0,
                                             (long)j.B * j.out * 4, stream));
                    hipLaunchKernelGGL(k_mfma_fwd2,
                                       dim3(ntm * ntn * ksplit), dim3(256),
                                       0, stream, j.x1, j.x2, j.wt,
                                       part_scratch, j.B, j.in1, j.in2,
                                       j.out, ksplit);
                    hipLaunchKernelGGL(k_mfma_epilogue, dim3(1024),
                                       dim3(256), 0, stream, part_scratch,
                                       j.bias, (const float*)nullptr, j.y,
                                       (long)j.B * j.out, j.out, j.act, 0);
                    continue;
                }
                int ntm = ceil_div(j.B, MT_M), ntn = ceil_div(j.out, MT_N);
                hipLaunchKernelGGL(k_mfma_fwd, dim3(ntm * ntn), dim3(256),
                                   0, stream, j.x1, j.x2, j.wt, j.bias, j.y,
                                   j.B, j.in1, j.in2, j.out, j.act, 1,
                                   (float*)nullptr);
            }
            return;
        }
        for (auto& j : jobs) {
            a[n++] = j;
            wgs = j.wg0 + j.nwg_b * j.nwg_o;
        }
        hipLaunchKernelGGL(k_fwd3, dim3(wgs), dim3(256), 0, stream,
                           a[0], a[1], a[2], n);
    }

    void launch_bwd(const float* dz, const float* x1, const float* x2,
                    const float* slab, float* gslab, const LayerDesc& l,
                    float* dx1, float* dx2, const float* h1, int prev_act,
                    bool want_dw) {
        int in_total = l.in1 + l.in2;
        if (cfg.batch >= 512) {
            if (want_dw) {
                // split K (= batch) until the grid covers the chip;
                // segments write disjoint partials, k_dw_reduce sums
                // (atomic epilogue retired — see k_dw_reduce comment).
                // Concat layers (in2 > 0) run as TWO GEMMs: the x1 block
                // with fully vectorized staging, and the tiny x2 (action)
                // block separately — one mixed GEMM left its edge tiles
                // on the guarded scalar path, and those straggler wgs
                // held the whole 200-us dispatch (profiles/wide_dissect2)
                auto dw_gemm = [&](const float* xop, int in_n, float* dwt,
                                   float* db) {
                    int ntm = ceil_div(in_n, MT_M);
                    int ntn = ceil_div(l.out, MT_N);
                    int ksplit = 1;
                    while (ntm * ntn * ksplit < 256 &&
                           ksplit * 2 * MT_K <= cfg.batch)
                        ksplit *= 2;
                    hipLaunchKernelGGL(k_mfma_dw, dim3(ntm * ntn * ksplit),
                                       dim3(256), 0, stream, dz, xop,
                                       (const float*)nullptr, dwt, db,
                                       cfg.batch, in_n, 0, l.out, ksplit,
                                       dw_parts);
                    if (ksplit > 1)
                        hipLaunchKernelGGL(k_dw_reduce, dim3(1024),
                                           dim3(256), 0, stream, dw_parts,
                                           dwt, db, (long)in_n * l.out,
                                           l.out, ksplit);
                };
                dw_gemm(x1, l.in1, gslab + l.w_off, gslab + l.b_off);
                if (l.in2 > 0)
                    dw_gemm(x2, l.in2,
                            gslab + l.w_off + (long)l.in1 * l.out,
                            (float*)nullptr);
            }
            if (dx1) {
                int ntm = ceil_div(cfg.batch, MT_M);
                int ntn = ceil_div(l.in1, MT_N);
                hipLaunchKernelGGL(k_mfma_dx, dim3(ntm * ntn), dim3(256),
                                   0, stream, dz, slab + l.w_off, h1, dx1,
                                   cfg.batch, 0, l.in1, l.out, prev_act);
            }
            if (dx2) {
                if (l.in2 <= 8) {
                    // action slice: wave-per-row VALU beats a 1-N-tile
                    // MFMA grid by ~7x here (see k_dx_narrow comment)
                    hipLaunchKernelGGL(k_dx_narrow, dim3(256), dim3(256),
                                       0, stream, dz, slab + l.w_off,
                                       (const float*)nullptr, dx2,
                                       cfg.batch, l.in1, in_total, l.out,
                                       ACT_NONE);
                } else {
                    int ntm = ceil_div(cfg.batch, MT_M);
                    int ntn = ceil_div(l.in2, MT_N);
                    hipLaunchKernelGGL(k_mfma_dx, dim3(ntm * ntn),
                                       dim3(256), 0, stream, dz,
                                       slab + l.w_off,
                                       (const float*)nullptr, dx2,
                                       cfg.batch, l.in1, in_total, l.out,
                                       ACT_NONE);
                }
            }
            return;
        }
        BwdJob j{};
        j.dz = dz; j.x1 = x1; j.x2 = x2;
        j.wt = slab + l.w_off;
        j.dwt = want_dw ? gslab + l.w_off : nullptr;
        j.dbias = want_dw ? gslab + l.b_off : nullptr;
        j.dx1 = dx1; j.dx2 = dx2; j.h1 = h1;
        j.B = cfg.batch; j.in1 = l.in1; j.in2 = l.in2; j.out = l.out;
        j.prev_act = prev_act;
        int wg = 0;
        if (want_dw) {
            j.wg0_dw = 0;
            j.nwg_dw_i = ceil_div(l.in1 + l.in2, BWT);
            j.nwg_dw_o = ceil_div(l.out, BWT);
            wg += j.nwg_dw_i * j.nwg_dw_o;
        }
        j.wg0_dx = wg;
        if (dx1 || dx2) {
            j.nwg_dx_b = ceil_div(cfg.batch, BXB);
            j.nwg_dx_i = ceil_div(l.in1 + l.in2, BXI);
            wg += j.nwg_dx_b * j.nwg_dx_i;
        } else { j.nwg_dx_b = j.nwg_dx_i = 0; }
        hipLaunchKernelGGL(k_bwd, dim3(wg), dim3(256), 0, stream, j);
    }

    // ---------------- the train step (one launch sequence) ----------------
    bool use_persistent() const {
        // the persistent megakernel assumes: probe/row waves cover the batch
        // (B <= PNWG*4), fan-ins fit the LDS x-stage, softmax <= one wave,
        // AND all PNWG workgroups co-resident (the software grid barrier
        // deadlocks otherwise — persistent_fits_ is the ctor's occupancy
        // check; on an occupancy-limited device we fall back to the
        // row-block + hipGraph path instead of corrupting state)
        return persistent_fits_ && cfg.batch <= 256 &&
               cfg.obs <= PXMAX && cfg.hidden + cfg.act <= PXMAX &&
               cfg.hidden <= PXMAX && cfg.atoms <= 64;
    }

    PStepArgs ps_args() {
        PStepArgs g{};
        g.B = cfg.batch; g.O = cfg.obs; g.A = cfg.act; g.H = cfg.hidden;
        g.K = cfg.atoms;
        g.v_min = cfg.v_min; g.v_max = cfg.v_max; g.gamma_n = cfg.gamma_n;
        g.per_eps = cfg.per_eps; g.tau = cfg.tau;
        g.lr_actor = cfg.lr_actor; g.lr_critic = cfg.lr_critic;
        g.per_alpha = cfg.per_alpha; g.per_beta0 = cfg.per_beta0;
        g.per_beta_iters = (float)cfg.per_beta_iters;
        g.is_weighting = cfg.is_weighting;
        g.seed = cfg.seed; g.tree_cap = tree_cap;
        g.n_actor = anet.n_params; g.n_critic = cnet.n_params;
        for (int i = 0; i < 4; ++i) { g.al[i] = anet.l[i]; g.cl[i] = cnet.l[i]; }
        g.p_actor = p_actor; g.p_actor_t = p_actor_t;
        g.p_critic = p_critic; g.p_critic_t = p_critic_t;
        g.g_actor = g_actor; g.g_critic = g_critic;
        g.m_actor = m_actor; g.v_actor = v_actor;
        g.m_critic = m_critic; g.v_critic = v_critic;
        g.rs = rs; g.ra = ra; g.rr = rr; g.rs2 = rs2; g.rd = rd;
        g.sum_tree = sum_tree; g.min_tree = min_tree;
        g.bs = bs; g.bs_b = bs_b;
        g.ba = ba; g.br = br; g.bs2 = bs2; g.bd = bd; g.bw = bw;
        g.pri = pri; g.bidx = bidx;
        g.at_h1 = at_h1; g.at_h2 = at_h2; g.at_h3 = at_h3; g.a2 = a2;
        g.ct_h1 = ct_h1; g.ct_h2 = ct_h2; g.ct_h3 = ct_h3; g.p_t = p_t;
        g.m_proj = m_proj;
        g.c_h1 = c_h1; g.c_h2 = c_h2; g.c_h3 = c_h3; g.q = q;
        g.dlog = dlog; g.d3 = d3; g.d2 = d2; g.d1 = d1;
        g.pa_h1 = pa_h1; g.pa_h2 = pa_h2; g.pa_h3 = pa_h3; g.a_out = a_out;
        g.pc_h1 = pc_h1; g.pc_h2 = pc_h2; g.pc_h3 = pc_h3; g.pq = pq;
        g.pd3 = pd3; g.pd2 = pd2; g.pdh1 = pdh1; g.adz = adz;
        g.az1 = az1; g.az2 = az2; g.az3 = az3;
        g.cnt = cnt; g.gbar = gbar; g.tstamp = tstamp;
        return g;
    }

    bool use_chain() const {
        // chain-fused variant: three parallel PR-row chains must fit the
        // grid, H must be a 64-multiple (LDS row padding), heads <= a wave.
        // Measured SLOWER than the layer-parallel phase kernel (1596 vs
        // 2274 steps/s at B=64 — the 10-layer serial policy megachain
        // dominates), so opt-in via D4PG_CHAIN=1 for further tuning.
        const char* e = getenv("D4PG_CHAIN");
        if (!e || e[0] != '1') return false;
        return use_persistent() && chain_fits_ && cfg.hidden % 64 == 0 &&
               3 * ceil_div(cfg.batch, PR) <= PNWG && cfg.act <= 64;
    }

    void enqueue_persistent(int nsteps) {
        PStepArgs g = ps_args();
        if (use_chain())
            hipLaunchKernelGGL(k_step_chain, dim3(PNWG), dim3(256), 0,
                               stream, g, nsteps);
        else
            hipLaunchKernelGGL(k_step_persistent, dim3(PNWG), dim3(256), 0,
                               stream, g, nsteps);
    }

    bool use_row_block() const {
        // row-block wastes weight bandwidth at large batch (every wave
        // re-streams every weight); the per-layer tiled path wins there.
        return cfg.batch <= 512;
    }

    RowBlockArgs rb_args() {
        RowBlockArgs g{};
        g.B = cfg.batch; g.O = cfg.obs; g.A = cfg.act; g.H = cfg.hidden;
        g.K = cfg.atoms;
        g.v_min = cfg.v_min; g.v_max = cfg.v_max; g.gamma_n = cfg.gamma_n;
        g.per_eps = cfg.per_eps; g.is_weighting = cfg.is_weighting;
        g.p_actor = p_actor; g.p_actor_t = p_actor_t;
        g.p_critic = p_critic; g.p_critic_t = p_critic_t;
        for (int i = 0; i < 4; ++i) { g.al[i] = anet.l[i]; g.cl[i] = cnet.l[i]; }
        g.bs = bs; g.ba = ba; g.br = br; g.bs2 = bs2; g.bd = bd; g.bw = bw;
        g.a2 = a2; g.p_t = p_t; g.m_proj = m_proj; g.q = q; g.dlog = dlog;
        g.pri = pri;
        g.c_h1 = c_h1; g.c_h2 = c_h2; g.c_h3 = c_h3;
        g.d1 = d1; g.d2 = d2; g.d3 = d3;
        g.pa_h1 = pa_h1; g.pa_h2 = pa_h2; g.pa_h3 = pa_h3; g.a_out = a_out;
        g.pc_h1 = pc_h1; g.pc_h2 = pc_h2; g.pc_h3 = pc_h3; g.pq = pq;
        g.az1 = az1; g.az2 = az2; g.az3 = az3; g.adz = adz;
        g.cnt = cnt;
        return g;
    }

    // dW-only job for one layer (used by the batched k_bwd4 launch)
    BwdJob dw_job(const float* dz, const float* x1, const float* x2,
                  const float* slab, float* gslab, const LayerDesc& l,
                  int& wg) {
        BwdJob j{};
        j.dz = dz; j.x1 = x1; j.x2 = x2;
        j.wt = slab + l.w_off;
        j.dwt = gslab + l.w_off; j.dbias = gslab + l.b_off;
        j.B = cfg.batch; j.in1 = l.in1; j.in2 = l.in2; j.out = l.out;
        j.wg0_dw = wg;
        j.nwg_dw_i = ceil_div(l.in1 + l.in2, BWT);
        j.nwg_dw_o = ceil_div(l.out, BWT);
        wg += j.nwg_dw_i * j.nwg_dw_o;
        j.wg0_dx = wg; j.nwg_dx_b = j.nwg_dx_i = 0;
        return j;
    }

    // Phase mask for split stepping (learner data parallelism: the host
    // all-reduces g_critic / g_actor between the GRADS and APPLY phases,
    // SURVEY §2c collectives list).  PH_ALL == a whole train step.
    enum { PH_CRITIC_GRADS = 1, PH_CRITIC_APPLY = 2,
           PH_ACTOR_GRADS = 4, PH_ACTOR_APPLY = 8, PH_ALL = 15 };

    void enqueue_step_rowblock(int mask = PH_ALL) {
        const int B = cfg.batch, H = cfg.hidden;
        int row_wgs = ceil_div(B, 4);
        long smem = rb_lds_bytes(H);

        RowBlockArgs g = rb_args();
        if (mask & PH_CRITIC_GRADS) {
        hipLaunchKernelGGL(k_per_sample, dim3(row_wgs), dim3(256), 0, stream,
            sum_tree, min_tree, tree_cap, rs, ra, rr, rs2, rd,
            cfg.obs, cfg.act, bs, ba, br, bs2, bd, bw, bidx, B, cnt,
            cfg.per_beta0, (float)cfg.per_beta_iters, cfg.seed);

        hipLaunchKernelGGL(k_critic_rowblock, dim3(row_wgs), dim3(256),
                           smem, stream, g);
        {
            int wg = 0;
            BwdJob j0 = dw_job(d1, bs, nullptr, p_critic, g_critic,
                               cnet.l[0], wg);
            BwdJob j1 = dw_job(d2, c_h1, ba, p_critic, g_critic,
                               cnet.l[1], wg);
            BwdJob j2 = dw_job(d3, c_h2, nullptr, p_critic, g_critic,
                               cnet.l[2], wg);
            BwdJob j3 = dw_job(dlog, c_h3, nullptr, p_critic, g_critic,
                               cnet.l[3], wg);
            hipLaunchKernelGGL(k_bwd4, dim3(wg), dim3(256), 0, stream,
                               j0, j1, j2, j3);
        }
        }
        if (mask & PH_CRITIC_APPLY)
        hipLaunchKernelGGL(k_adam_lerp, dim3(256), dim3(256), 0, stream,
                           p_critic, g_critic, m_critic, v_critic,
                           p_critic_t, cnet.n_params, cfg.lr_critic,
                           0.9f, 0.999f, 1e-8f, cfg.tau, cnt, 0);
        if (mask & PH_ACTOR_GRADS) {
        hipLaunchKernelGGL(k_policy_rowblock, dim3(row_wgs), dim3(256),
                           smem, stream, g);
        {
            int wg = 0;
            BwdJob j0 = dw_job(az1, bs, nullptr, p_actor, g_actor,
                               anet.l[0], wg);
            BwdJob j1 = dw_job(az2, pa_h1, nullptr, p_actor, g_actor,
                               anet.l[1], wg);
            BwdJob j2 = dw_job(az3, pa_h2, nullptr, p_actor, g_actor,
                               anet.l[2], wg);
            BwdJob j3 = dw_job(adz, pa_h3, nullptr, p_actor, g_actor,
                               anet.l[3], wg);
            hipLaunchKernelGGL(k_bwd4, dim3(wg), dim3(256), 0, stream,
                               j0, j1, j2, j3);
        }
        }
        if (mask & PH_ACTOR_APPLY) {
        hipLaunchKernelGGL(k_adam_lerp, dim3(256), dim3(256), 0, stream,
                           p_actor, g_actor, m_actor, v_actor,
                           p_actor_t, anet.n_params, cfg.lr_actor,
                           0.9f, 0.999f, 1e-8f, cfg.tau, cnt, 1);
        hipLaunchKernelGGL(k_per_update, dim3(1), dim3(256), 0, stream,
                           sum_tree, min_tree, tree_cap, bidx, pri, B,
                           cfg.per_alpha, cnt);
        }
    }

    void enqueue_step(int mask = PH_ALL) {
        if (use_persistent() && mask == PH_ALL) {
            enqueue_persistent(1); return;
        }
        if (use_row_block()) { enqueue_step_rowblock(mask); return; }
        const int B = cfg.batch, K = cfg.atoms, H = cfg.hidden;
        const int waves_per_wg = 4;
        int row_wgs = ceil_div(B, waves_per_wg);

        if (mask & PH_CRITIC_GRADS) {
        hipLaunchKernelGGL(k_per_sample, dim3(row_wgs), dim3(256), 0, stream,
            sum_tree, min_tree, tree_cap, rs, ra, rr, rs2, rd,
            cfg.obs, cfg.act, bs, ba, br, bs2, bd, bw, bidx, B, cnt,
            cfg.per_beta0, (float)cfg.per_beta_iters, cfg.seed);

        // P2: a_t.L1(s2) | c_t.L1(s2) | c.L1(s)
        {
            int wg = 0;
            auto j0 = fwd_job(bs2, nullptr, p_actor_t, anet.l[0], at_h1,
                              ACT_RELU, wg);
            auto j1 = fwd_job(bs2, nullptr, p_critic_t, cnet.l[0], ct_h1,
                              ACT_RELU, wg);
            auto j2 = fwd_job(bs, nullptr, p_critic, cnet.l[0], c_h1,
                              ACT_RELU, wg);
            launch_fwd({j0, j1, j2});
        }
        // P3: a_t.L2 | c.L2(cat c_h1, a)
        {
            int wg = 0;
            auto j0 = fwd_job(at_h1, nullptr, p_actor_t, anet.l[1], at_h2,
                              ACT_NONE, wg);
            auto j1 = fwd_job(c_h1, ba, p_critic, cnet.l[1], c_h2,
                              ACT_RELU, wg);
            launch_fwd({j0, j1});
        }
        // P4: a_t.L3 | c.L3
        {
            int wg = 0;
            auto j0 = fwd_job(at_h2, nullptr, p_actor_t, anet.l[2], at_h3,
                              ACT_RELU, wg);
            auto j1 = fwd_job(c_h2, nullptr, p_critic, cnet.l[2], c_h3,
                              ACT_RELU, wg);
            launch_fwd({j0, j1});
        }
        // P5: a_t.L4 tanh | c.L4 softmax -> q
        {
            int wg = 0;
            auto j0 = fwd_job(at_h3, nullptr, p_actor_t, anet.l[3], a2,
                              ACT_TANH, wg);
            auto j1 = fwd_job(c_h3, nullptr, p_critic, cnet.l[3], q,
                              ACT_SOFTMAX, wg);
            launch_fwd({j0, j1});
        }
        // P6-8: c_t.L2(cat ct_h1, a2), c_t.L3, c_t.L4 softmax -> p_t
        { int wg = 0; launch_fwd({fwd_job(ct_h1, a2, p_critic_t, cnet.l[1],
                                          ct_h2, ACT_RELU, wg)}); }
        { int wg = 0; launch_fwd({fwd_job(ct_h2, nullptr, p_critic_t,
                                          cnet.l[2], ct_h3, ACT_RELU, wg)}); }
        { int wg = 0; launch_fwd({fwd_job(ct_h3, nullptr, p_critic_t,
                                          cnet.l[3], p_t, ACT_SOFTMAX, wg)}); }
        // P9+P10 fused: projection + CE grad + priorities
        hipLaunchKernelGGL(k_project_ce, dim3(min(row_wgs, 256)),
                           dim3(256),
                           waves_per_wg * 64 * sizeof(float), stream,
                           p_t, br, bd, q, bw, m_proj, dlog, pri, cnt,
                           B, K, cfg.v_min, cfg.v_max, cfg.gamma_n,
                           cfg.per_eps, cfg.is_weighting);
        // P11-14: critic backward L4..L1
        launch_bwd(dlog, c_h3, nullptr, p_critic, g_critic, cnet.l[3],
                   d3, nullptr, c_h3, ACT_RELU, true);
        launch_bwd(d3, c_h2, nullptr, p_critic, g_critic, cnet.l[2],
                   d2, nullptr, c_h2, ACT_RELU, true);
        launch_bwd(d2, c_h1, ba, p_critic, g_critic, cnet.l[1],
                   d1, nullptr, c_h1, ACT_RELU, true);
        launch_bwd(d1, bs, nullptr, p_critic, g_critic, cnet.l[0],
                   nullptr, nullptr, nullptr, ACT_NONE, true);
        }
        // P15: Adam critic + target soft-update fused
        if (mask & PH_CRITIC_APPLY)
        hipLaunchKernelGGL(k_adam_lerp, dim3(256), dim3(256), 0, stream,
                           p_critic, g_critic, m_critic, v_critic,
                           p_critic_t, cnet.n_params, cfg.lr_critic,
                           0.9f, 0.999f, 1e-8f, cfg.tau, cnt, 0);
        if (mask & PH_ACTOR_GRADS) {
        // P16: actor.L1(s) | critic'.L1(s)
        {
            int wg = 0;
            auto j0 = fwd_job(bs, nullptr, p_actor, anet.l[0], pa_h1,
                              ACT_RELU, wg);
            auto j1 = fwd_job(bs, nullptr, p_critic, cnet.l[0], pc_h1,
                              ACT_RELU, wg);
            launch_fwd({j0, j1});
        }
        // P17-19: actor L2, L3, L4
        { int wg = 0; launch_fwd({fwd_job(pa_h1, nullptr, p_actor, anet.l[1],
                                          pa_h2, ACT_NONE, wg)}); }
        { int wg = 0; launch_fwd({fwd_job(pa_h2, nullptr, p_actor, anet.l[2],
                                          pa_h3, ACT_RELU, wg)}); }
        { int wg = 0; launch_fwd({fwd_job(pa_h3, nullptr, p_actor, anet.l[3],
                                          a_out, ACT_TANH, wg)}); }
        // P20-22: critic'(s, a_out)
        { int wg = 0; launch_fwd({fwd_job(pc_h1, a_out, p_critic, cnet.l[1],
                                          pc_h2, ACT_RELU, wg)}); }
        { int wg = 0; launch_fwd({fwd_job(pc_h2, nullptr, p_critic,
                                          cnet.l[2], pc_h3, ACT_RELU, wg)}); }
        { int wg = 0; launch_fwd({fwd_job(pc_h3, nullptr, p_critic,
                                          cnet.l[3], pq, ACT_SOFTMAX, wg)}); }
        // P23: policy head gradient
        hipLaunchKernelGGL(k_policy_grad, dim3(min(row_wgs, 256)),
                           dim3(256), 0, stream,
                           pq, pd3, cnt, B, K, cfg.v_min, cfg.v_max);
        // P24-26: dX back through critic' (no dW), ending at da
        launch_bwd(pd3, pc_h3, nullptr, p_critic, nullptr, cnet.l[3],
                   pd2, nullptr, pc_h3, ACT_RELU, false);
        launch_bwd(pd2, pc_h2, nullptr, p_critic, nullptr, cnet.l[2],
                   pdh1, nullptr, pc_h2, ACT_RELU, false);
        launch_bwd(pdh1, pc_h1, a_out, p_critic, nullptr, cnet.l[1],
                   nullptr, pda, pc_h1, ACT_RELU, false);
        // P27: tanh backward at actor output
        hipLaunchKernelGGL(k_tanh_bwd, dim3(ceil_div((long)B * cfg.act, 256)),
                           dim3(256), 0, stream, pda, a_out, adz,
                           (long)B * cfg.act);
        // P28-31: actor backward L4..L1 (with dW)
        launch_bwd(adz, pa_h3, nullptr, p_actor, g_actor, anet.l[3],
                   pd2 /*reuse*/, nullptr, pa_h3, ACT_RELU, true);
        launch_bwd(pd2, pa_h2, nullptr, p_actor, g_actor, anet.l[2],
                   pdh1, nullptr, pa_h2, ACT_NONE, true);
        launch_bwd(pdh1, pa_h1, nullptr, p_actor, g_actor, anet.l[1],
                   pd2, nullptr, pa_h1, ACT_RELU, true);
        launch_bwd(pd2, bs, nullptr, p_actor, g_actor, anet.l[0],
                   nullptr, nullptr, nullptr, ACT_NONE, true);
        }
        if (!(mask & PH_ACTOR_APPLY)) return;
        // P32: Adam actor + target soft-update fused (critic target was
        // lerped in P15; ordering matches the row-block/persistent paths)
        hipLaunchKernelGGL(k_adam_lerp, dim3(256), dim3(256), 0, stream,
                           p_actor, g_actor, m_actor, v_actor,
                           p_actor_t, anet.n_params, cfg.lr_actor,
                           0.9f, 0.999f, 1e-8f, cfg.tau, cnt, 1);
        // P34: PER priority write-back.  Large batches use the per-level
        // grid-wide repair (the one-wg level-synced kernel serializes).
        if (B >= 512) {
            hipLaunchKernelGGL(k_per_leaves, dim3(ceil_div(B, 256)),
                               dim3(256), 0, stream, sum_tree, min_tree,
                               tree_cap, bidx, pri, B, cfg.per_alpha, cnt);
            long levels = 0;
            for (long c = tree_cap; c > 1; c >>= 1) ++levels;
            // 4 heights per dispatch (k_per_level4): 20 launches -> 5
            for (long h0 = 0; h0 < levels; h0 += 4) {
                int nlv = (int)((levels - h0) < 4 ? (levels - h0) : 4);
                hipLaunchKernelGGL(k_per_level4,
                                   dim3(ceil_div((long)B * nlv, 256)),
                                   dim3(256), 0, stream, sum_tree, min_tree,
                                   tree_cap, bidx, B, h0, nlv);
            }
            hipLaunchKernelGGL(k_tick_end, dim3(1), dim3(64), 0, stream,
                               cnt);
        } else {
            hipLaunchKernelGGL(k_per_update, dim3(1), dim3(256), 0, stream,
                               sum_tree, min_tree, tree_cap, bidx, pri, B,
                               cfg.per_alpha, cnt);
        }
    }

    void step(int n) {
        bool persistent = use_persistent() && n > 0;
        if (persistent) {
            // whole multi-step run in ONE launch — zero host dispatch
            // between steps (numerically identical to n single launches)
            enqueue_persistent(n);
        } else {
            for (int i = 0; i < n; ++i) enqueue_step();
        }
        HIP_CHECK(hipStreamSynchronize(stream));
        if (persistent) check_bar_error();
    }

    // Partial step for learner data parallelism: run only the phases in
    // `mask` (PH_* above) and synchronize, so the host can all-reduce the
    // gradient slabs between GRADS and APPLY.  Never takes the persistent
    // megakernel (it fuses all phases); flagship shapes go through the
    // row-block path, wide shapes through the per-layer MFMA path.
    void step_part(int mask) {
        enqueue_step(mask);
        HIP_CHECK(hipStreamSynchronize(stream));
    }

    void invalidate_graph() {
        if (graph_exec) { hipGraphExecDestroy(graph_exec); graph_exec = nullptr; }
        if (graph) { hipGraphDestroy(graph); graph = nullptr; }
        graph_steps = 0;
    }

    void capture(int steps_per_graph) {
        invalidate_graph();
        HIP_CHECK(hipStreamBeginCapture(stream, hipStreamCaptureModeThreadLocal));
        for (int i = 0; i < steps_per_graph; ++i) enqueue_step();
        HIP_CHECK(hipStreamEndCapture(stream, &graph));
        HIP_CHECK(hipGraphInstantiate(&graph_exec, graph, nullptr, nullptr, 0));
        graph_steps = steps_per_graph;
    }

    void replay(int iters) {
        if (!graph_exec) throw std::runtime_error("no graph captured");
        for (int i = 0; i < iters; ++i)
            HIP_CHECK(hipGraphLaunch(graph_exec, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
    }

    // non-blocking variants for overlap with host work
    void replay_async(int iters) {
        for (int i = 0; i < iters; ++i)
            HIP_CHECK(hipGraphLaunch(graph_exec, stream));
    }
    void sync() { HIP_CHECK(hipStreamSynchronize(stream)); }

    // ------------- replay ops -------------
    void synth_fill(long n, uint64_t seed) {
        if (n > cfg.capacity) n = cfg.capacity;
        hipLaunchKernelGGL(k_synth_fill, dim3(1024), dim3(256), 0, stream,
                           rs, ra, rr, rs2, rd, sum_tree, min_tree, tree_cap,
                           cfg.capacity, n, cfg.obs, cfg.act, seed,
                           cfg.per_alpha);
        for (long lo = tree_cap / 2; lo >= 1; lo /= 2)
            hipLaunchKernelGGL(k_tree_build_level, dim3(256), dim3(256), 0,
                               stream, sum_tree, min_tree, lo,
                               lo == 0 ? 1 : 2 * lo);
        Counters h{};
        HIP_CHECK(hipMemcpyAsync(&h, cnt, sizeof(h), hipMemcpyDeviceToHost,
                                 stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        h.size = n; h.pos = n % cfg.capacity; h.max_priority = 1.0f;
        HIP_CHECK(hipMemcpy(cnt, &h, sizeof(h), hipMemcpyHostToDevice));
    }

    void ingest(const float* hs, const float* ha, const float* hr,
                const float* hs2, const float* hd, int T) {
        if (T > ing_cap) throw std::runtime_error("ingest batch too large");
        const int O = cfg.obs, A = cfg.act;
        HIP_CHECK(hipMemcpyAsync(ing_s, hs, (long)T * O * 4,
                                 hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(ing_a, ha, (long)T * A * 4,
                                 hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(ing_r, hr, (long)T * 4,
                                 hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(ing_s2, hs2, (long)T * O * 4,
                                 hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipMemcpyAsync(ing_d, hd, (long)T * 4,
                                 hipMemcpyHostToDevice, stream));
        if (T <= 4096) {
            hipLaunchKernelGGL(k_replay_add, dim3(1), dim3(1024), 0, stream,
                               rs, ra, rr, rs2, rd, sum_tree, min_tree,
                               tree_cap, cfg.capacity, ing_s, ing_a, ing_r,
                               ing_s2, ing_d, T, O, A, cfg.per_alpha, cnt);
        } else {
            // bulk path: grid-wide copy + leaf init, then a full
            // bandwidth-bound tree rebuild (far cheaper than one wg
            // chasing 20-level paths for 100k rows)
            Counters h{};
            HIP_CHECK(hipMemcpyAsync(&h, cnt, sizeof(h),
                                     hipMemcpyDeviceToHost, stream));
            HIP_CHECK(hipStreamSynchronize(stream));
            double pa = pow((double)h.max_priority, (double)cfg.per_alpha);
            hipLaunchKernelGGL(k_replay_copy, dim3(1024), dim3(256), 0,
                               stream, rs, ra, rr, rs2, rd, sum_tree,
                               min_tree, tree_cap, cfg.capacity, h.pos,
                               ing_s, ing_a, ing_r, ing_s2, ing_d, T, O, A,
                               pa);
            for (long lo = tree_cap / 2; lo >= 1; lo >>= 1)
                hipLaunchKernelGGL(k_tree_build_level, dim3(1024),
                                   dim3(256), 0, stream, sum_tree,
                                   min_tree, lo, 2 * lo);
            h.pos = (h.pos + T) % cfg.capacity;
            h.size = h.size + T > cfg.capacity ? cfg.capacity : h.size + T;
            HIP_CHECK(hipMemcpyAsync(cnt, &h, sizeof(h),
                                     hipMemcpyHostToDevice, stream));
        }
        HIP_CHECK(hipStreamSynchronize(stream));
    }

    // ------------- param I/O (host pointers, already transposed [in][out]) --
    void load_slab(float* dst, const float* src, long n) {
        HIP_CHECK(hipMemcpy(dst, src, n * 4, hipMemcpyHostToDevice));
    }
    void store_slab(float* dst, const float* src, long n) {
        HIP_CHECK(hipMemcpy(dst, src, n * 4, hipMemcpyDeviceToHost));
    }

    Counters read_counters() {
        Counters h{};
        HIP_CHECK(hipMemcpy(&h, cnt, sizeof(h), hipMemcpyDeviceToHost));
        // adam/rng counters are pre-advanced for the NEXT step (init 1 so
        // the first step sees t=1); report steps COMPLETED to the host.
        // beta_t starts at 0 (LinearSchedule value_at(0) on the first
        // sample) and ticks at step end, so its raw value IS steps done.
        h.adam_t_actor -= 1;
        h.adam_t_critic -= 1;
        h.rng_epoch -= 1;
        return h;
    }

    // ---------------- GPU-resident rollout (device env + K11 + K15) -----
    // See the kernel-section comment above k_roll_begin.  State lives in a
    // separate pool so rollout capacity is independent of the learner
    // batch; the actor weights and the replay are SHARED with the learner
    // (same p_actor slab, same rs/../sum_tree), which is the point: the
    // whole actor->replay->learner loop stays in HBM.
    RollArgs roll_{};
    void* roll_pool_ = nullptr;
    float *r_h1 = nullptr, *r_h2 = nullptr, *r_h3 = nullptr;
    hipGraph_t roll_graph = nullptr;
    hipGraphExec_t roll_graph_exec = nullptr;
    int roll_emits_per_ep = 0;

    void rollout_free() {
        if (roll_graph_exec) { hipGraphExecDestroy(roll_graph_exec);
                               roll_graph_exec = nullptr; }
        if (roll_graph) { hipGraphDestroy(roll_graph); roll_graph = nullptr; }
        if (roll_pool_) { hipFree(roll_pool_); roll_pool_ = nullptr; }
    }

    void rollout_alloc(int M, int nsteps, int horizon, float gamma,
                       int noise_kind, float eps, float ou_theta,
                       float ou_sigma, float ou_mu, uint64_t seed) {
        if (cfg.obs != 3 || cfg.act != 1)
            throw std::runtime_error(
                "device rollout models Pendulum dynamics (obs=3, act=1)");
        if (horizon < nsteps)
            throw std::runtime_error("rollout horizon < n_steps");
        rollout_free();
        RollArgs a{};
        a.M = M; a.O = 3; a.A = 1; a.n = nsteps; a.horizon = horizon;
        a.gamma = gamma; a.eps = eps; a.ou_theta = ou_theta;
        a.ou_sigma = ou_sigma; a.ou_mu = ou_mu; a.ou_dt = 1e-2f;
        a.noise_kind = noise_kind; a.seed = seed;
        a.per_alpha = cfg.per_alpha;
        const int H = cfg.hidden;
        long off = 0;
        auto sub = [&](long nelem) {
            long o = off;
            off += (nelem * 4 + 255) & ~255L;
            return o;
        };
        long o_th = sub(M), o_td = sub(M), o_obs = sub((long)M * 3),
             o_act = sub(M), o_ou = sub(M),
             o_sr = sub((long)nsteps * M * 3), o_ar = sub((long)nsteps * M),
             o_rr = sub((long)nsteps * M), o_ep = sub(4),
             o_h1 = sub((long)M * H), o_h2 = sub((long)M * H),
             o_h3 = sub((long)M * H);
        HIP_CHECK(hipMalloc(&roll_pool_, off));
        HIP_CHECK(hipMemset(roll_pool_, 0, off));
        char* base = (char*)roll_pool_;
        a.th = (float*)(base + o_th); a.thdot = (float*)(base + o_td);
        a.obs = (float*)(base + o_obs); a.act = (float*)(base + o_act);
        a.ou_x = (float*)(base + o_ou);
        a.s_ring = (float*)(base + o_sr); a.a_ring = (float*)(base + o_ar);
        a.r_ring = (float*)(base + o_rr);
        a.ep_state = (long long*)(base + o_ep);
        r_h1 = (float*)(base + o_h1); r_h2 = (float*)(base + o_h2);
        r_h3 = (float*)(base + o_h3);
        a.rs = rs; a.ra = ra; a.rr = rr; a.rs2 = rs2; a.rd = rd;
        a.sum_tree = sum_tree; a.min_tree = min_tree;
        a.tree_cap = tree_cap; a.capacity = cfg.capacity;
        a.cnt = cnt;
        roll_ = a;
        roll_emits_per_ep = horizon - nsteps + 1;
    }

    void roll_fwd_chain() {
        const int M = roll_.M;
        auto run = [&](const float* x, const LayerDesc& l, float* y,
                       int actk) {
            if (M >= 512) {
                int ntm = ceil_div(M, MT_M), ntn = ceil_div(l.out, MT_N);
                hipLaunchKernelGGL(k_mfma_fwd, dim3(ntm * ntn), dim3(256),
                                   0, stream, x, (const float*)nullptr,
                                   p_actor + l.w_off, p_actor + l.b_off, y,
                                   M, l.in1, l.in2, l.out, actk, 1,
                                   (float*)nullptr);
            } else {
                FwdJob j{};
                j.x1 = x; j.x2 = nullptr;
                j.wt = p_actor + l.w_off; j.bias = p_actor + l.b_off;
                j.y = y; j.B = M; j.in1 = l.in1; j.in2 = l.in2;
                j.out = l.out; j.act = actk;
                j.wg0 = 0; j.nwg_b = ceil_div(M, TB);
                j.nwg_o = ceil_div(l.out, TO);
                FwdJob e{};
                hipLaunchKernelGGL(k_fwd3, dim3(j.nwg_b * j.nwg_o),
                                   dim3(256), 0, stream, j, e, e, 1);
            }
        };
        run(roll_.obs, anet.l[0], r_h1, ACT_RELU);
        run(r_h1, anet.l[1], r_h2, ACT_NONE);   // fc2->fc2_2 no-act quirk
        run(r_h2, anet.l[2], r_h3, ACT_RELU);
        run(r_h3, anet.l[3], roll_.act, ACT_TANH);
    }

    void rollout_enqueue_episode(bool reset) {
        RollArgs& a = roll_;
        int wgs = ceil_div(a.M, 256);
        hipLaunchKernelGGL(k_roll_begin, dim3(1), dim3(64), 0, stream, a);
        if (reset)
            hipLaunchKernelGGL(k_roll_reset, dim3(wgs), dim3(256), 0,
                               stream, a);
        int emit_k = 0;
        for (int t = 0; t < a.horizon; ++t) {
            roll_fwd_chain();
            int ek = (t >= a.n - 1) ? emit_k : -1;
            int done = (t == a.horizon - 1) ? 1 : 0;
            hipLaunchKernelGGL(k_roll_tick, dim3(wgs), dim3(256), 0, stream,
                               a, t, t % a.n, ek, done);
            if (ek >= 0) ++emit_k;
        }
        hipLaunchKernelGGL(k_roll_end, dim3(1), dim3(64), 0, stream, a,
                           (long)emit_k * a.M);
        // internal tree nodes: one bandwidth-bound full sweep per episode
        // (vs horizon x per-path repair) — leaves were set at emit
        for (long lo = tree_cap / 2; lo >= 1; lo >>= 1)
            hipLaunchKernelGGL(k_tree_build_level, dim3(1024), dim3(256), 0,
                               stream, sum_tree, min_tree, lo, 2 * lo);
    }

    void rollout_run(int episodes, bool reset, bool use_graph) {
        if (!roll_pool_)
            throw std::runtime_error("rollout_alloc first");
        if (use_graph && reset) {
            if (!roll_graph_exec) {
                HIP_CHECK(hipStreamBeginCapture(
                    stream, hipStreamCaptureModeThreadLocal));
                rollout_enqueue_episode(true);
                HIP_CHECK(hipStreamEndCapture(stream, &roll_graph));
                HIP_CHECK(hipGraphInstantiate(&roll_graph_exec, roll_graph,
                                              nullptr, nullptr, 0));
            }
            for (int e = 0; e < episodes; ++e)
                HIP_CHECK(hipGraphLaunch(roll_graph_exec, stream));
        } else {
            for (int e = 0; e < episodes; ++e)
                rollout_enqueue_episode(reset);
        }
        HIP_CHECK(hipStreamSynchronize(stream));
    }

    // test hook: inject exact env state (parity vs the numpy oracle)
    void rollout_set_state(const float* th, const float* td, int M) {
        if (!roll_pool_ || M != roll_.M)
            throw std::runtime_error("rollout_set_state: bad M");
        HIP_CHECK(hipMemcpy(roll_.th, th, M * 4, hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(roll_.thdot, td, M * 4, hipMemcpyHostToDevice));
        std::vector<float> obs(3 * M);
        for (int i = 0; i < M; ++i) {
            obs[i * 3 + 0] = cosf(th[i]);
            obs[i * 3 + 1] = sinf(th[i]);
            obs[i * 3 + 2] = td[i];
        }
        HIP_CHECK(hipMemcpy(roll_.obs, obs.data(), 3L * M * 4,
                            hipMemcpyHostToDevice));
        HIP_CHECK(hipMemset(roll_.ou_x, 0, M * 4));
    }

    // actor forward for eval/smoke: x[B,obs] (device via staging) -> a[B,act]
    void actor_forward(const float* hx, float* hy, int n) {
        // reuse batch buffers (requires n <= batch)
        if (n > cfg.batch) throw std::runtime_error("actor_forward: n > batch");
        HIP_CHECK(hipMemcpyAsync(bs, hx, (long)n * cfg.obs * 4,
                                 hipMemcpyHostToDevice, stream));
        int wg;
        wg = 0; launch_fwd({fwd_job(bs, nullptr, p_actor, anet.l[0], pa_h1,
                                    ACT_RELU, wg)});
        wg = 0; launch_fwd({fwd_job(pa_h1, nullptr, p_actor, anet.l[1], pa_h2,
                                    ACT_NONE, wg)});
        wg = 0; launch_fwd({fwd_job(pa_h2, nullptr, p_actor, anet.l[2], pa_h3,
                                    ACT_RELU, wg)});
        wg = 0; launch_fwd({fwd_job(pa_h3, nullptr, p_actor, anet.l[3], a_out,
                                    ACT_TANH, wg)});
        HIP_CHECK(hipMemcpyAsync(hy, a_out, (long)n * cfg.act * 4,
                                 hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
    }
};

}  // namespace d4pg
