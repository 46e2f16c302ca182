// gossipy_amd CDNA4 (gfx950) kernels — the batched compute path of the
// MI355X gossip engine.
//
// Design: one kernel launch services EVERY simulated node
// active in a tick sub-phase. Workgroups map 1:1 to nodes; a node's whole
// event sequence for the sub-phase (merge -> local SGD -> reply snapshot)
// runs inside its workgroup so no cross-workgroup ordering is ever needed —
// ordering between sub-phases comes from stream order of the launches.
//
// Kernel inventory (K-numbers from SURVEY.md §2.4):
//   snapshot_kernel       — K-equivalent of ModelHandler.caching
//                           (gossipy/model/handler.py:160-176): arena row
//                           copy instead of copy.deepcopy.
//   tick_logreg_kernel    — K3+K4+K5 fused: per-receiver merge (mean of
//                           state, gossipy/model/handler.py:260-280) +
//                           minibatch SGD on CE(sigmoid(Wx+b))
//                           (gossipy/model/handler.py:235-258,
//                            gossipy/model/nn.py:162-166).
//   tick_linear_kernel    — K1/K2: Pegasos hinge SGD
//                           (gossipy/model/handler.py:416-423) and AdaLine
//                           delta rule (gossipy/model/handler.py:364-368),
//                           one wave per node, weights in registers.
//   tick_mlp_kernel       — K3/K4 generalized: fused MLP fwd/bwd/SGD
//                           (gossipy/model/nn.py:91-99).
//
// Models are tiny (57x2 logreg ... few-hundred-wide MLPs); per-launch work
// is launch-latency bound, so kernels are built to let ONE launch cover all
// active nodes and to keep every intermediate in LDS/registers — the D
// parameters are read from HBM once per tick and written once.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_cooperative_groups.h>

#define DEV_INLINE __device__ __forceinline__

constexpr int KMAX = 16;     // max classes for the logreg path
constexpr int WAVE = 64;

// CreateModelMode numbering shared with engine/backend.py (_MODE_ID)
constexpr int MODE_UPDATE = 0;
constexpr int MODE_MERGE_UPDATE = 1;
constexpr int MODE_UPDATE_MERGE = 2;
constexpr int MODE_PASS = 3;

DEV_INLINE float wave_sum(float v)
{
    for (int off = WAVE / 2; off > 0; off >>= 1)
        v += __shfl_down(v, off, WAVE);
    return __shfl(v, 0, WAVE);
}

// reduction when only the first <=8 lanes carry data (e.g. the MF rank-k
// dot, k<=8): 3 shuffle steps instead of 6, broadcast to every lane
DEV_INLINE float wave_sum8(float v)
{
    v += __shfl_down(v, 4, WAVE);
    v += __shfl_down(v, 2, WAVE);
    v += __shfl_down(v, 1, WAVE);
    return __shfl(v, 0, WAVE);
}


// ---------------------------------------------------------------------------
// snapshot: slots[slot_ids[i]] = params[nodes[i]]  (+ age)
// ---------------------------------------------------------------------------

__global__ void snapshot_kernel(
    const float* __restrict__ params,
    const int* __restrict__ ages,
    float* __restrict__ slots,
    int* __restrict__ slot_ages,
    const int* __restrict__ nodes,
    const int* __restrict__ slot_ids,
    int n, int W, int Dp, int src_off, int A)
    // W = slot width, Dp = full row width, src_off = sub-block offset
    // (MF snapshots only the item block), A = age width
{
    long total = (long)n * W;
    for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
         idx += (long)gridDim.x * blockDim.x) {
        int row = idx / W;
        int col = idx - (long)row * W;
        int node = nodes[row];
        int slot = slot_ids[row];
        slots[(long)slot * W + col] = params[(long)node * Dp + src_off + col];
        if (col < A) slot_ages[(long)slot * A + col] = ages[(long)node * A + col];
    }
}

// ---------------------------------------------------------------------------
// logreg tick: one workgroup per active node
// ---------------------------------------------------------------------------

struct LogregArgs {
    float* params; int* ages;
    float* slots; int* slot_ages;
    const int* nodes; const int* ptr;
    const int* dslots; const int* rslots;
    const int* dmodes;  // per-delivery pass-through flag (1 = PASS), or null
    const float* X; const float* Y; const int* counts;
    int d, k, Smax, D;
    float lr, wd;
    int epochs, bs, mode, update_only;
};

// Minibatch SGD epochs over the node's shard, on the LDS-resident model W.
// xb/dz are LDS scratch; age is a wave-uniform register.
DEV_INLINE void logreg_update(const LogregArgs& a, int node, float* W,
                              float* xb, float* dz, int& age,
                              int prestaged = 0)
{
    int tid = threadIdx.x;
    int c = a.counts[node];
    if (c == 0) return;
    int bsz = (a.bs == 0) ? c : min(a.bs, c);
    const float* Xn = a.X + (long)node * a.Smax * a.d;
    const float* Yn = a.Y + (long)node * a.Smax;
    // single-batch shards (every BASELINE config): xb stays valid across
    // epochs AND deliveries, so it is staged at most once per receiver row
    int xb_valid = prestaged;
    for (int ep = 0; ep < a.epochs; ++ep) {
        for (int s0 = 0; s0 < c; s0 += bsz) {
            int m = min(bsz, c - s0);
            // stage the batch in LDS (coalesced: consecutive threads read
            // consecutive floats of the shard); the caller may have staged
            // the very first batch already (fused with the W load)
            if (!xb_valid) {
                for (int e = tid; e < m * a.d; e += blockDim.x)
                    xb[e] = Xn[(long)s0 * a.d + e];
            }
            xb_valid = (c <= bsz);
            __syncthreads();
            // per-sample forward + dLoss/dz (thread = sample)
            if (tid < m) {
                float z[KMAX];
                for (int kk = 0; kk < a.k; ++kk) {
                    float acc = W[a.k * a.d + kk];  // bias
                    const float* wrow = W + kk * a.d;
                    const float* xrow = xb + tid * a.d;
                    for (int dd = 0; dd < a.d; ++dd) acc += wrow[dd] * xrow[dd];
                    z[kk] = acc;
                }
                // a = sigmoid(z); p = softmax(a); dz = (p - 1_y)/m * a*(1-a)
                float amax = -1e30f;
                for (int kk = 0; kk < a.k; ++kk) {
                    z[kk] = 1.0f / (1.0f + __expf(-z[kk]));
                    amax = fmaxf(amax, z[kk]);
                }
                float sum = 0.f;
                float p[KMAX];
                for (int kk = 0; kk < a.k; ++kk) {
                    p[kk] = __expf(z[kk] - amax);
                    sum += p[kk];
                }
                int yi = (int)Yn[s0 + tid];
                float inv = 1.0f / sum;
                for (int kk = 0; kk < a.k; ++kk) {
                    float g = p[kk] * inv - (kk == yi ? 1.0f : 0.0f);
                    dz[tid * a.k + kk] = (g / m) * z[kk] * (1.0f - z[kk]);
                }
            }
            __syncthreads();
            // SGD step (thread = parameter)
            for (int e = tid; e < a.k * a.d; e += blockDim.x) {
                int kk = e / a.d, dd = e - kk * a.d;
                float g = 0.f;
                for (int s = 0; s < m; ++s) g += dz[s * a.k + kk] * xb[s * a.d + dd];
                if (a.wd != 0.f) g += a.wd * W[e];
                W[e] -= a.lr * g;
            }
            for (int e = tid; e < a.k; e += blockDim.x) {
                float g = 0.f;
                for (int s = 0; s < m; ++s) g += dz[s * a.k + e];
                W[a.k * a.d + e] -= a.lr * g;
            }
            __syncthreads();
            age += 1;  // wave-uniform (gossipy/model/handler.py:258)
        }
    }
}

DEV_INLINE void logreg_process_node(const LogregArgs& a, int i)
{
    int node = a.nodes[i];
    int tid = threadIdx.x;
    extern __shared__ float sm[];
    float* W = sm;                         // D
    float* W2 = W + a.D;                   // D (UPDATE_MERGE scratch)
    float* xb = W2 + a.D;                  // bsmax*d
    int bsmax = (a.bs == 0) ? a.Smax : min(a.bs, a.Smax);
    float* dz = xb + bsmax * a.d;          // bsmax*k

    // Fast path for the common MERGE_UPDATE single-sequence case: fuse the
    // model load with the FIRST delivery's merge and prefetch the first
    // minibatch in the same phase — three independent HBM streams behind
    // one barrier instead of three serial chains.
    int j0 = a.update_only ? 0 : a.ptr[i];
    int j1 = a.update_only ? 0 : a.ptr[i + 1];
    if (!a.update_only && j1 > j0 && a.mode == MODE_MERGE_UPDATE &&
        !(a.dmodes && a.dmodes[j0])) {
        int slot = a.dslots[j0];
        const float* srow = a.slots + (long)slot * a.D;
        for (int e = tid; e < a.D; e += blockDim.x)
            W[e] = 0.5f * (a.params[(long)node * a.D + e] + srow[e]);
        int c = a.counts[node];
        int m0 = min((a.bs == 0) ? max(c, 1) : a.bs, c);
        for (int e = tid; e < m0 * a.d; e += blockDim.x)
            xb[e] = a.X[(long)node * a.Smax * a.d + e];
        __syncthreads();
        int age = max(a.ages[node], a.slot_ages[slot]);
        // xb holds batch 0 after the fused stage above; it STAYS valid for
        // later deliveries only when the shard is a single batch
        int c0 = a.counts[node];
        int xb_keep = (a.bs == 0) || (c0 <= a.bs);
        // register double-buffer: issue the NEXT delivery's slot-row loads
        // before each (long) update so the HBM latency hides under compute
        constexpr int PRE = 8;
        float pre[PRE];
        int pre_age = 0;
        int npre = min(a.D, PRE * (int)blockDim.x);
        auto issue_pre = [&](int j) {
            if (j < j1 && !(a.dmodes && a.dmodes[j])) {
                const float* srow2 = a.slots + (long)a.dslots[j] * a.D;
                for (int u = 0; u < PRE; ++u) {
                    int e = tid + u * (int)blockDim.x;
                    pre[u] = (e < a.D) ? srow2[e] : 0.f;
                }
                pre_age = a.slot_ages[a.dslots[j]];
            }
        };
        issue_pre(j0 + 1);
        logreg_update(a, node, W, xb, dz, age, /*prestaged=*/1);
        int rs0 = a.rslots ? a.rslots[j0] : -1;
        if (rs0 >= 0) {  // first delivery's PUSH_PULL reply snapshot
            for (int e = tid; e < a.D; e += blockDim.x)
                a.slots[(long)rs0 * a.D + e] = W[e];
            if (tid == 0) a.slot_ages[rs0] = age;
            __syncthreads();
        }
        for (int j = j0 + 1; j < j1; ++j) {
            int slot2 = a.dslots[j];
            const float* srow2 = a.slots + (long)slot2 * a.D;
            int md = (a.dmodes && a.dmodes[j]) ? MODE_PASS : a.mode;
            if (md == MODE_MERGE_UPDATE) {
                for (int u = 0; u < PRE; ++u) {
                    int e = tid + u * (int)blockDim.x;
                    if (e < a.D) W[e] = 0.5f * (W[e] + pre[u]);
                }
                for (int e = npre + tid; e < a.D; e += blockDim.x)
                    W[e] = 0.5f * (W[e] + srow2[e]);
                age = max(age, pre_age);
                __syncthreads();
                issue_pre(j + 1);
                logreg_update(a, node, W, xb, dz, age, /*prestaged=*/xb_keep);
            } else {  // PASS (pass-through coin resolved to adopt)
                for (int e = tid; e < a.D; e += blockDim.x) W[e] = srow2[e];
                age = a.slot_ages[slot2];
                __syncthreads();
                issue_pre(j + 1);
            }
            int rs2 = a.rslots ? a.rslots[j] : -1;
            if (rs2 >= 0) {
                for (int e = tid; e < a.D; e += blockDim.x)
                    a.slots[(long)rs2 * a.D + e] = W[e];
                if (tid == 0) a.slot_ages[rs2] = age;
                __syncthreads();
            }
        }
        __syncthreads();
        for (int e = tid; e < a.D; e += blockDim.x)
            a.params[(long)node * a.D + e] = W[e];
        if (tid == 0) a.ages[node] = age;
        return;
    }

    for (int e = tid; e < a.D; e += blockDim.x)
        W[e] = a.params[(long)node * a.D + e];
    __syncthreads();
    int age = a.ages[node];

    if (a.update_only) {
        logreg_update(a, node, W, xb, dz, age);
    } else {
        for (int j = a.ptr[i]; j < a.ptr[i + 1]; ++j) {
            int slot = a.dslots[j];
            const float* srow = a.slots + (long)slot * a.D;
            int sage = a.slot_ages[slot];
            // pass-through gossip resolves some deliveries to PASS
            // (gossipy/node.py:380-392)
            int md = (a.dmodes && a.dmodes[j]) ? MODE_PASS : a.mode;
            if (md == MODE_MERGE_UPDATE) {
                for (int e = tid; e < a.D; e += blockDim.x)
                    W[e] = 0.5f * (W[e] + srow[e]);
                age = max(age, sage);
                __syncthreads();
                logreg_update(a, node, W, xb, dz, age);
            } else if (md == MODE_UPDATE) {
                // adopt the received model, then train it
                for (int e = tid; e < a.D; e += blockDim.x) W[e] = srow[e];
                age = sage;
                __syncthreads();
                logreg_update(a, node, W, xb, dz, age);
            } else if (md == MODE_UPDATE_MERGE) {
                logreg_update(a, node, W, xb, dz, age);
                for (int e = tid; e < a.D; e += blockDim.x) W2[e] = srow[e];
                __syncthreads();
                int age2 = sage;
                logreg_update(a, node, W2, xb, dz, age2);
                for (int e = tid; e < a.D; e += blockDim.x)
                    W[e] = 0.5f * (W[e] + W2[e]);
                age = max(age, age2);
                __syncthreads();
            } else {  // PASS
                for (int e = tid; e < a.D; e += blockDim.x) W[e] = srow[e];
                age = sage;
                __syncthreads();
            }
            // reply-delivery launches pass rslots == nullptr (replies to
            // replies are discarded, gossipy/simul.py:426)
            int rs = a.rslots ? a.rslots[j] : -1;
            if (rs >= 0) {  // PUSH_PULL reply snapshot (post merge+update)
                for (int e = tid; e < a.D; e += blockDim.x)
                    a.slots[(long)rs * a.D + e] = W[e];
                if (tid == 0) a.slot_ages[rs] = age;
                __syncthreads();
            }
        }
    }
    __syncthreads();
    for (int e = tid; e < a.D; e += blockDim.x)
        a.params[(long)node * a.D + e] = W[e];
    if (tid == 0) a.ages[node] = age;
}

__global__ void __launch_bounds__(128)
tick_logreg_kernel(LogregArgs a)
{
    logreg_process_node(a, blockIdx.x);
}

// ---------------------------------------------------------------------------
// all2all weighted k-way merge (K5 weighted variant; WeightedTMH._merge,
// gossipy/model/handler.py:666-688): params[n] = w0*params[n] + sum_j
// w_j*slots[s_j]; age = max over merged. Family-agnostic (elementwise over
// D); the per-family update-only kernel runs right after it.
// ---------------------------------------------------------------------------

__global__ void wmerge_kernel(
    float* __restrict__ params,
    int* __restrict__ ages,
    const float* __restrict__ slots,
    const int* __restrict__ slot_ages,
    const int* __restrict__ nodes,
    const int* __restrict__ ptr,
    const int* __restrict__ wslots,
    const float* __restrict__ wweights,
    const float* __restrict__ selfw,
    int D)
{
    int i = blockIdx.x;
    int node = nodes[i];
    int tid = threadIdx.x;
    int lo = ptr[i], hi = ptr[i + 1];
    float w0 = selfw[i];
    for (int e = tid; e < D; e += blockDim.x) {
        float acc = w0 * params[(long)node * D + e];
        for (int j = lo; j < hi; ++j)
            acc += wweights[j] * slots[(long)wslots[j] * D + e];
        params[(long)node * D + e] = acc;
    }
    if (tid == 0) {
        int age = ages[node];
        for (int j = lo; j < hi; ++j) age = max(age, slot_ages[wslots[j]]);
        ages[node] = age;
    }
}

// ---------------------------------------------------------------------------
// partitioned logreg tick (K7/K8, PartitionedTMH semantics)
//
// Parity: gossipy/model/handler.py:455-525 + gossipy/model/sampling.py:
// 201-234. Each node keeps a per-partition age vector; a delivery merges
// ONE partition (age-weighted), and each local batch increments every
// partition age then divides each parameter's gradient by its partition's
// age. The partition cover is the reference's F-flat equal split mapped to
// arena offsets via the device-resident perm/pptr arrays (engine/models.py
// _PartitionMixin).
// ---------------------------------------------------------------------------

struct LogregPartArgs {
    float* params; int* ages;            // ages: [n_local, P]
    float* slots; int* slot_ages;        // slot_ages: [cap, P]
    const int* nodes; const int* ptr;
    const int* dslots; const int* rslots; const int* dpids;
    const float* X; const float* Y; const int* counts;
    const int* perm;   // [D]  partition-ordered -> arena offset
    const int* pptr;   // [P+1]
    const int* apart;  // [D]  arena offset -> partition id
    int P, d, k, Smax, D;
    float lr, wd;
    int epochs, bs, mode, update_only;
    //: xb holds the WHOLE shard (staged once per receiver row) instead of
    //: one re-staged minibatch slice per step per delivery — decided
    //: host-side from the LDS budget. The reference math is unchanged;
    //: the staging was the tokenized row's repeated HBM round trip.
    int stage_full;
};

// Minibatch SGD with whole-age-vector increment per batch and per-partition
// gradient rescale (gossipy/model/handler.py:503-520).
DEV_INLINE void logreg_part_update(const LogregPartArgs& a, int node, float* W,
                                   int* agev, float* xb, float* dz)
{
    int tid = threadIdx.x;
    int c = a.counts[node];
    if (c == 0) return;
    int bsz = (a.bs == 0) ? c : min(a.bs, c);
    const float* Xn = a.X + (long)node * a.Smax * a.d;
    const float* Yn = a.Y + (long)node * a.Smax;
    for (int ep = 0; ep < a.epochs; ++ep) {
        for (int s0 = 0; s0 < c; s0 += bsz) {
            int m = min(bsz, c - s0);
            float* xs = a.stage_full ? xb + (long)s0 * a.d : xb;
            if (!a.stage_full) {
                for (int e = tid; e < m * a.d; e += blockDim.x)
                    xb[e] = Xn[(long)s0 * a.d + e];
            }
            // self.n_updates += 1 happens before the step (handler.py:506)
            for (int p = tid; p < a.P; p += blockDim.x) agev[p] += 1;
            __syncthreads();
            if (tid < m) {
                float z[KMAX];
                for (int kk = 0; kk < a.k; ++kk) {
                    float acc = W[a.k * a.d + kk];
                    const float* wrow = W + kk * a.d;
                    const float* xrow = xs + tid * a.d;
                    for (int dd = 0; dd < a.d; ++dd) acc += wrow[dd] * xrow[dd];
                    z[kk] = acc;
                }
                float amax = -1e30f;
                for (int kk = 0; kk < a.k; ++kk) {
                    z[kk] = 1.0f / (1.0f + __expf(-z[kk]));
                    amax = fmaxf(amax, z[kk]);
                }
                float sum = 0.f;
                float p[KMAX];
                for (int kk = 0; kk < a.k; ++kk) {
                    p[kk] = __expf(z[kk] - amax);
                    sum += p[kk];
                }
                int yi = (int)Yn[s0 + tid];
                float inv = 1.0f / sum;
                for (int kk = 0; kk < a.k; ++kk) {
                    float g = p[kk] * inv - (kk == yi ? 1.0f : 0.0f);
                    dz[tid * a.k + kk] = (g / m) * z[kk] * (1.0f - z[kk]);
                }
            }
            __syncthreads();
            for (int e = tid; e < a.k * a.d; e += blockDim.x) {
                int kk = e / a.d, dd = e - kk * a.d;
                float g = 0.f;
                for (int s = 0; s < m; ++s) g += dz[s * a.k + kk] * xs[s * a.d + dd];
                g /= (float)agev[a.apart[e]];  // _adjust_gradient
                if (a.wd != 0.f) g += a.wd * W[e];
                W[e] -= a.lr * g;
            }
            for (int e = tid; e < a.k; e += blockDim.x) {
                float g = 0.f;
                for (int s = 0; s < m; ++s) g += dz[s * a.k + e];
                g /= (float)agev[a.apart[a.k * a.d + e]];
                if (a.wd != 0.f) g += a.wd * W[a.k * a.d + e];
                W[a.k * a.d + e] -= a.lr * g;
            }
            __syncthreads();
        }
    }
}

// Age-weighted merge of partition `pid` of src into W
// (handler.py:497-501; weights (0,0) -> (1,1)). src/src_ages may point to
// LDS (trained received model) or HBM (slot row).
DEV_INLINE void logreg_part_merge(const LogregPartArgs& a, float* W, int* agev,
                                  const float* src, const int* src_ages, int pid)
{
    int tid = threadIdx.x;
    int w1 = agev[pid], w2 = src_ages[pid];
    float m1, m2;
    if (w1 == 0 && w2 == 0) { m1 = 0.5f; m2 = 0.5f; }
    else {
        float s = (float)(w1 + w2);
        m1 = (float)w1 / s;
        m2 = (float)w2 / s;
    }
    __syncthreads();
    for (int e = a.pptr[pid] + tid; e < a.pptr[pid + 1]; e += blockDim.x) {
        int i = a.perm[e];
        W[i] = m1 * W[i] + m2 * src[i];
    }
    __syncthreads();
    if (tid == 0) agev[pid] = max(w1, w2);
    __syncthreads();
}

DEV_INLINE void logreg_part_process_node(const LogregPartArgs& a, int i)
{
    int node = a.nodes[i];
    int tid = threadIdx.x;
    extern __shared__ float sm[];
    float* W = sm;                         // D
    float* W2 = W + a.D;                   // D (received-model scratch)
    float* xb = W2 + a.D;                  // (stage_full ? Smax : bsmax)*d
    int bsmax = (a.bs == 0) ? a.Smax : min(a.bs, a.Smax);
    int xbn = a.stage_full ? a.Smax : bsmax;
    float* dz = xb + xbn * a.d;            // bsmax*k
    int* agev = (int*)(dz + bsmax * a.k);  // P
    int* agev2 = agev + a.P;               // P

    for (int e = tid; e < a.D; e += blockDim.x)
        W[e] = a.params[(long)node * a.D + e];
    for (int p = tid; p < a.P; p += blockDim.x)
        agev[p] = a.ages[(long)node * a.P + p];
    if (a.stage_full) {
        // whole shard -> LDS once; every delivery's every minibatch step
        // slices it (the re-staging was a repeated HBM round trip on the
        // tokenized row's sequential chains)
        int c = a.counts[node];
        const float* Xn = a.X + (long)node * a.Smax * a.d;
        for (int e = tid; e < c * a.d; e += blockDim.x) xb[e] = Xn[e];
    }
    __syncthreads();

    if (a.update_only) {
        logreg_part_update(a, node, W, agev, xb, dz);
    } else {
        for (int j = a.ptr[i]; j < a.ptr[i + 1]; ++j) {
            int slot = a.dslots[j];
            int pid = a.dpids ? a.dpids[j] : 0;
            const float* srow = a.slots + (long)slot * a.D;
            const int* sages = a.slot_ages + (long)slot * a.P;
            if (a.mode == MODE_MERGE_UPDATE) {
                logreg_part_merge(a, W, agev, srow, sages, pid);
                logreg_part_update(a, node, W, agev, xb, dz);
            } else {
                // UPDATE: train the received model, merge its partition
                // (handler.py:481-483). UPDATE_MERGE: also self-update first
                // (handler.py:487-490). PASS is rejected host-side.
                if (a.mode == MODE_UPDATE_MERGE)
                    logreg_part_update(a, node, W, agev, xb, dz);
                for (int e = tid; e < a.D; e += blockDim.x) W2[e] = srow[e];
                for (int p = tid; p < a.P; p += blockDim.x) agev2[p] = sages[p];
                __syncthreads();
                logreg_part_update(a, node, W2, agev2, xb, dz);
                logreg_part_merge(a, W, agev, W2, agev2, pid);
            }
            int rs = a.rslots ? a.rslots[j] : -1;
            if (rs >= 0) {
                for (int e = tid; e < a.D; e += blockDim.x)
                    a.slots[(long)rs * a.D + e] = W[e];
                for (int p = tid; p < a.P; p += blockDim.x)
                    a.slot_ages[(long)rs * a.P + p] = agev[p];
                __syncthreads();
            }
        }
    }
    __syncthreads();
    for (int e = tid; e < a.D; e += blockDim.x)
        a.params[(long)node * a.D + e] = W[e];
    for (int p = tid; p < a.P; p += blockDim.x)
        a.ages[(long)node * a.P + p] = agev[p];
}

__global__ void __launch_bounds__(128)
tick_logreg_part_kernel(LogregPartArgs a)
{
    logreg_part_process_node(a, blockIdx.x);
}

// ---------------------------------------------------------------------------
// sampled logreg tick (K6, SamplingTMH semantics)
//
// Parity: gossipy/model/handler.py:426-452 + gossipy/model/sampling.py:
// 75-107. Each delivery averages only a seeded random coordinate subset
// (with replacement); the index sequence is splitmix64(seed + j) % D —
// bit-identical to engine/rng.py sample_indices, so the torch oracle and
// this kernel draw the same coordinates. Ages are not merged.
// ---------------------------------------------------------------------------

DEV_INLINE unsigned long long splitmix64_dev(unsigned long long x)
{
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

struct LogregSampArgs {
    float* params; int* ages;
    float* slots; int* slot_ages;
    const int* nodes; const int* ptr;
    const int* dslots; const int* rslots; const int* dseeds;
    const float* X; const float* Y; const int* counts;
    int samp_c, d, k, Smax, D;
    float lr, wd;
    int epochs, bs, mode, update_only;
};

// W3 holds the pre-merge W (duplicate sampled indices must all read the
// ORIGINAL value — torch advanced-index assignment semantics).
DEV_INLINE void logreg_samp_merge(const LogregSampArgs& a, float* W, float* W3,
                                  const float* src, int seed)
{
    int tid = threadIdx.x;
    for (int e = tid; e < a.D; e += blockDim.x) W3[e] = W[e];
    __syncthreads();
    for (int q = tid; q < a.samp_c; q += blockDim.x) {
        int i = (int)(splitmix64_dev((unsigned long long)seed +
                                     (unsigned long long)q) %
                      (unsigned long long)a.D);
        W[i] = 0.5f * (W3[i] + src[i]);
    }
    __syncthreads();
}

__global__ void __launch_bounds__(128)
tick_logreg_samp_kernel(LogregSampArgs a)
{
    int i = blockIdx.x;
    int node = a.nodes[i];
    int tid = threadIdx.x;
    extern __shared__ float sm[];
    float* W = sm;                         // D
    float* W2 = W + a.D;                   // D (received-model scratch)
    float* W3 = W2 + a.D;                  // D (pre-merge snapshot)
    float* xb = W3 + a.D;                  // bsmax*d
    int bsmax = (a.bs == 0) ? a.Smax : min(a.bs, a.Smax);
    float* dz = xb + bsmax * a.d;          // bsmax*k

    for (int e = tid; e < a.D; e += blockDim.x)
        W[e] = a.params[(long)node * a.D + e];
    __syncthreads();
    int age = a.ages[node];

    // reuse the plain logreg update math: wrap args in a LogregArgs view
    LogregArgs u;
    u.X = a.X; u.Y = a.Y; u.counts = a.counts;
    u.d = a.d; u.k = a.k; u.Smax = a.Smax; u.D = a.D;
    u.lr = a.lr; u.wd = a.wd; u.epochs = a.epochs; u.bs = a.bs;

    if (a.update_only) {
        logreg_update(u, node, W, xb, dz, age);
    } else {
        for (int j = a.ptr[i]; j < a.ptr[i + 1]; ++j) {
            int slot = a.dslots[j];
            int seed = a.dseeds ? a.dseeds[j] : 0;
            const float* srow = a.slots + (long)slot * a.D;
            int sage = a.slot_ages[slot];
            if (a.mode == MODE_MERGE_UPDATE) {
                logreg_samp_merge(a, W, W3, srow, seed);
                logreg_update(u, node, W, xb, dz, age);
            } else {
                // UPDATE: train received, merge its sample into self
                // (handler.py:440-442); UPDATE_MERGE: self-update first
                // (handler.py:445-448)
                if (a.mode == MODE_UPDATE_MERGE)
                    logreg_update(u, node, W, xb, dz, age);
                for (int e = tid; e < a.D; e += blockDim.x) W2[e] = srow[e];
                __syncthreads();
                int age2 = sage;
                logreg_update(u, node, W2, xb, dz, age2);
                logreg_samp_merge(a, W, W3, W2, seed);
            }
            int rs = a.rslots ? a.rslots[j] : -1;
            if (rs >= 0) {
                for (int e = tid; e < a.D; e += blockDim.x)
                    a.slots[(long)rs * a.D + e] = W[e];
                if (tid == 0) a.slot_ages[rs] = age;
                __syncthreads();
            }
        }
    }
    __syncthreads();
    for (int e = tid; e < a.D; e += blockDim.x)
        a.params[(long)node * a.D + e] = W[e];
    if (tid == 0) a.ages[node] = age;
}

// ---------------------------------------------------------------------------
// matrix-factorization tick (K9/K10, MFModelHandler semantics)
//
// Parity: gossipy/model/handler.py:528-576. Per delivery: item-block
// age-weighted merge with the reference's extra /2 (n_updates NOT merged),
// then per-rating SGD where Y[i] uses the OLD X and X uses the NEW Y[i].
// One wavefront per receiver: lanes cover the k latent dims in the rating
// loop and stride the item block in the merge.
// ---------------------------------------------------------------------------

struct MFArgs {
    float* params; int* ages;
    float* slots; int* slot_ages;
    const int* nodes; const int* ptr;
    const int* dslots; const int* rslots;
    const float* X; const float* Y; const int* counts;  // items ride X[:,:,0]
    int k, n_items, Smax, D, item_off, Wslot;
    float reg, lr;
    int update_only;
};

DEV_INLINE void mf_update(const MFArgs& a, int node, float* row, int& age)
{
    // per-rating chain is order-dependent: wave 0 runs it; other waves of
    // the (wider) block idle here and rejoin at the caller's barrier.
    //
    // The chain is the MF round's critical path (551 µs/dispatch measured
    // at 10k nodes, 83% of round GPU time), so it is register-pipelined:
    // the user factors X and both biases live in REGISTERS for the whole
    // sweep (merges never touch them — they only rewrite the item block),
    // and the NEXT rating's item row + item bias are prefetched while the
    // current rating computes. A prefetch of the item just updated is
    // patched from registers (exact — the only write between prefetch and
    // use is the current item's).
    int lane = threadIdx.x;
    if (lane >= WAVE) return;
    int c = a.counts[node];
    if (c == 0) return;
    const float* items = a.X + (long)node * a.Smax;
    const float* ratings = a.Y + (long)node * a.Smax;
    float shrink = 1.0f - a.reg * a.lr;
    int coff = a.item_off + a.n_items * a.k;
    float x = (lane < a.k) ? row[lane] : 0.f;
    float b = row[a.k];
    int item = (int)items[0];
    float* Yi = row + a.item_off + (long)item * a.k;
    float yi = (lane < a.k) ? Yi[lane] : 0.f;
    float ci = row[coff + item];
    for (int s = 0; s < c; ++s) {
        int item_n = 0;
        float* Yi_n = nullptr;
        float yi_n = 0.f, ci_n = 0.f;
        if (s + 1 < c) {  // prefetch next item row + bias
            item_n = (int)items[s + 1];
            Yi_n = row + a.item_off + (long)item_n * a.k;
            yi_n = (lane < a.k) ? Yi_n[lane] : 0.f;
            ci_n = row[coff + item_n];
        }
        float r = ratings[s];
        float part = (lane < a.k) ? x * yi : 0.f;
        // k <= 8: 3-step reduction, bit-identical to the 64-lane tree
        // (the extra lanes only ever add zeros, so the pairings match)
        float dot = (a.k <= 8) ? wave_sum8(part) : wave_sum(part);
        float err = r - dot - b - ci;
        float yi_new = shrink * yi + a.lr * err * x;
        float ci_new = ci + a.lr * err;
        if (lane < a.k) {
            Yi[lane] = yi_new;
            x = shrink * x + a.lr * err * yi_new;
        }
        if (lane == 0) row[coff + item] = ci_new;
        b += a.lr * err;
        age += 1;
        if (s + 1 < c) {
            if (item_n == item) {  // re-rated item: fix up from registers
                yi_n = yi_new;
                ci_n = ci_new;
            }
            item = item_n;
            Yi = Yi_n;
            yi = yi_n;
            ci = ci_n;
        }
    }
    if (lane < a.k) row[lane] = x;
    if (lane == 0) row[a.k] = b;
}

DEV_INLINE void mf_merge(const MFArgs& a, float* row, int age,
                         const float* srow, int sage)
{
    float den = 2.0f * (age + sage);
    float w1 = (float)age / den, w2 = (float)sage / den;
    for (int e = threadIdx.x; e < a.Wslot; e += blockDim.x)
        row[a.item_off + e] = w1 * row[a.item_off + e] + w2 * srow[e];
}

__global__ void __launch_bounds__(256)
tick_mf_kernel(MFArgs a)
{
    // 256 threads: the item-block merge/copy loops (tens of KB per row)
    // use the full block; the order-dependent rating chain runs on wave 0
    int i = blockIdx.x;
    int node = a.nodes[i];
    float* row = a.params + (long)node * a.D;
    int age = a.ages[node];
    if (a.update_only) {
        mf_update(a, node, row, age);
        __syncthreads();
    } else {
        __shared__ int age_sh;
        for (int j = a.ptr[i]; j < a.ptr[i + 1]; ++j) {
            int slot = a.dslots[j];
            mf_merge(a, row, age, a.slots + (long)slot * a.Wslot,
                     a.slot_ages[slot]);
            __syncthreads();
            mf_update(a, node, row, age);
            __syncthreads();
            // the rating chain ran on wave 0 only — rebroadcast the age so
            // a second delivery's merge weights agree across all waves
            if (threadIdx.x == 0) age_sh = age;
            __syncthreads();
            age = age_sh;
            int rs = a.rslots ? a.rslots[j] : -1;
            if (rs >= 0) {
                for (int e = threadIdx.x; e < a.Wslot; e += blockDim.x)
                    a.slots[(long)rs * a.Wslot + e] = row[a.item_off + e];
                // thread 0 (wave 0) holds the post-update age
                if (threadIdx.x == 0) a.slot_ages[rs] = age;
                __syncthreads();
            }
        }
    }
    if (threadIdx.x == 0) a.ages[node] = age;
}

// ---------------------------------------------------------------------------
// k-means tick (K11, KMeansHandler semantics, naive matching)
//
// Parity: gossipy/model/handler.py:579-639. Per update call: every sample
// assigns to its nearest centroid; the EMA applies with torch
// indexed-assignment semantics — per centroid only the LAST assigned
// sample lands (handler.py:613-615); n_updates bumps once per call and is
// never merged. Centroids live in LDS.
// ---------------------------------------------------------------------------

struct KMeansArgs {
    float* params; int* ages;
    float* slots; int* slot_ages;
    const int* nodes; const int* ptr;
    const int* dslots; const int* rslots;
    const float* X; const int* counts;
    int k, dim, Smax, D;
    float alpha;
    int mode, update_only;
};

DEV_INLINE void kmeans_update(const KMeansArgs& a, int node, float* C,
                              int* assign, int& age)
{
    int tid = threadIdx.x;
    int c = a.counts[node];
    if (c == 0) return;
    const float* Xn = a.X + (long)node * a.Smax * a.dim;
    // assignment: thread = sample
    for (int s = tid; s < c; s += blockDim.x) {
        const float* x = Xn + (long)s * a.dim;
        float best = 1e30f;
        int bj = 0;
        for (int j = 0; j < a.k; ++j) {
            const float* cj = C + j * a.dim;
            float d2 = 0.f;
            for (int e = 0; e < a.dim; ++e) {
                float t = x[e] - cj[e];
                d2 += t * t;
            }
            if (d2 < best) { best = d2; bj = j; }
        }
        assign[s] = bj;
    }
    __syncthreads();
    // EMA, last-write-wins: thread = centroid, scan for its last sample
    for (int j = tid; j < a.k; j += blockDim.x) {
        int last = -1;
        for (int s = 0; s < c; ++s)
            if (assign[s] == j) last = s;
        if (last >= 0) {
            const float* x = Xn + (long)last * a.dim;
            float* cj = C + j * a.dim;
            for (int e = 0; e < a.dim; ++e)
                cj[e] = cj[e] * (1.0f - a.alpha) + a.alpha * x[e];
        }
    }
    __syncthreads();
    age += 1;
}

__global__ void __launch_bounds__(128)
tick_kmeans_kernel(KMeansArgs a)
{
    int i = blockIdx.x;
    int node = a.nodes[i];
    int tid = threadIdx.x;
    extern __shared__ float sm[];
    float* C = sm;                       // D = k*dim
    float* C2 = C + a.D;                 // received-model scratch
    int* assign = (int*)(C2 + a.D);      // Smax

    for (int e = tid; e < a.D; e += blockDim.x)
        C[e] = a.params[(long)node * a.D + e];
    __syncthreads();
    int age = a.ages[node];

    if (a.update_only) {
        kmeans_update(a, node, C, assign, age);
    } else {
        for (int j = a.ptr[i]; j < a.ptr[i + 1]; ++j) {
            int slot = a.dslots[j];
            const float* srow = a.slots + (long)slot * a.D;
            int sage = a.slot_ages[slot];
            if (a.mode == MODE_MERGE_UPDATE) {
                for (int e = tid; e < a.D; e += blockDim.x)
                    C[e] = 0.5f * (C[e] + srow[e]);
                __syncthreads();
                kmeans_update(a, node, C, assign, age);
            } else if (a.mode == MODE_UPDATE) {
                for (int e = tid; e < a.D; e += blockDim.x) C[e] = srow[e];
                age = sage;
                __syncthreads();
                kmeans_update(a, node, C, assign, age);
            } else if (a.mode == MODE_UPDATE_MERGE) {
                kmeans_update(a, node, C, assign, age);
                for (int e = tid; e < a.D; e += blockDim.x) C2[e] = srow[e];
                __syncthreads();
                int age2 = sage;
                kmeans_update(a, node, C2, assign, age2);
                for (int e = tid; e < a.D; e += blockDim.x)
                    C[e] = 0.5f * (C[e] + C2[e]);
                __syncthreads();
            } else {  // PASS
                for (int e = tid; e < a.D; e += blockDim.x) C[e] = srow[e];
                age = sage;
                __syncthreads();
            }
            int rs = a.rslots ? a.rslots[j] : -1;
            if (rs >= 0) {
                for (int e = tid; e < a.D; e += blockDim.x)
                    a.slots[(long)rs * a.D + e] = C[e];
                if (tid == 0) a.slot_ages[rs] = age;
                __syncthreads();
            }
        }
    }
    __syncthreads();
    for (int e = tid; e < a.D; e += blockDim.x)
        a.params[(long)node * a.D + e] = C[e];
    if (tid == 0) a.ages[node] = age;
}

// ---------------------------------------------------------------------------
// pegasos / adaline tick: one WAVE per node, weights in registers
// ---------------------------------------------------------------------------

struct LinearArgs {
    float* params; int* ages;
    float* slots; int* slot_ages;
    const int* nodes; const int* ptr;
    const int* dslots; const int* rslots;
    const int* dmodes;
    const float* X; const float* Y; const int* counts;
    int d, Smax;
    float lrlam;       // pegasos lambda or adaline lr
    int is_pegasos, mode, update_only;
};

constexpr int LIN_MAX_REGS = 16;  // supports d up to 16*64 = 1024

// Per-sample sequential pass (the math is order-dependent: Pegasos' step
// size is 1/(t*lam) with t = running age). w lives in per-lane registers
// (lane L owns dims L, L+64, ...).
DEV_INLINE void linear_update(const LinearArgs& a, int node, float* w,
                              int nreg, int& age)
{
    int lane = threadIdx.x;
    int c = a.counts[node];
    const float* Xn = a.X + (long)node * a.Smax * a.d;
    const float* Yn = a.Y + (long)node * a.Smax;
    for (int s = 0; s < c; ++s) {
        const float* x = Xn + (long)s * a.d;
        float part = 0.f;
        float xr[LIN_MAX_REGS];
        for (int r = 0; r < nreg; ++r) {
            int e = lane + r * WAVE;
            xr[r] = (e < a.d) ? x[e] : 0.f;
            part += w[r] * xr[r];
        }
        float pred = wave_sum(part);
        float y = Yn[s];
        if (a.is_pegasos) {
            age += 1;
            float lr = 1.0f / (age * a.lrlam);
            float hinge = (pred * y - 1.0f < 0.f) ? 1.0f : 0.0f;
            float scale = 1.0f - lr * a.lrlam;
            float add = hinge * lr * y;
            for (int r = 0; r < nreg; ++r) w[r] = w[r] * scale + add * xr[r];
        } else {
            float err = y - pred;
            for (int r = 0; r < nreg; ++r) w[r] += a.lrlam * err * xr[r];
        }
    }
    if (!a.is_pegasos) age += c;  // AdaLine ages by sample count
}

DEV_INLINE void linear_process_node(const LinearArgs& a, int i)
{
    int node = a.nodes[i];
    int lane = threadIdx.x & (WAVE - 1);
    int nreg = (a.d + WAVE - 1) / WAVE;
    float w[LIN_MAX_REGS];
    for (int r = 0; r < nreg; ++r) {
        int e = lane + r * WAVE;
        w[r] = (e < a.d) ? a.params[(long)node * a.d + e] : 0.f;
    }
    int age = a.ages[node];

    if (a.update_only) {
        linear_update(a, node, w, nreg, age);
    } else {
        for (int j = a.ptr[i]; j < a.ptr[i + 1]; ++j) {
            int slot = a.dslots[j];
            const float* srow = a.slots + (long)slot * a.d;
            int sage = a.slot_ages[slot];
            int md = (a.dmodes && a.dmodes[j]) ? MODE_PASS : a.mode;
            if (md == MODE_MERGE_UPDATE) {
                for (int r = 0; r < nreg; ++r) {
                    int e = lane + r * WAVE;
                    if (e < a.d) w[r] = 0.5f * (w[r] + srow[e]);
                }
                age = max(age, sage);
                linear_update(a, node, w, nreg, age);
            } else if (md == MODE_UPDATE) {
                for (int r = 0; r < nreg; ++r) {
                    int e = lane + r * WAVE;
                    if (e < a.d) w[r] = srow[e];
                }
                age = sage;
                linear_update(a, node, w, nreg, age);
            } else if (md == MODE_UPDATE_MERGE) {
                linear_update(a, node, w, nreg, age);
                float w2[LIN_MAX_REGS];
                for (int r = 0; r < nreg; ++r) {
                    int e = lane + r * WAVE;
                    w2[r] = (e < a.d) ? srow[e] : 0.f;
                }
                int age2 = sage;
                linear_update(a, node, w2, nreg, age2);
                for (int r = 0; r < nreg; ++r) w[r] = 0.5f * (w[r] + w2[r]);
                age = max(age, age2);
            } else {  // PASS
                for (int r = 0; r < nreg; ++r) {
                    int e = lane + r * WAVE;
                    if (e < a.d) w[r] = srow[e];
                }
                age = sage;
            }
            int rs = a.rslots ? a.rslots[j] : -1;
            if (rs >= 0) {
                for (int r = 0; r < nreg; ++r) {
                    int e = lane + r * WAVE;
                    if (e < a.d) a.slots[(long)rs * a.d + e] = w[r];
                }
                if (lane == 0) a.slot_ages[rs] = age;
            }
        }
    }
    for (int r = 0; r < nreg; ++r) {
        int e = lane + r * WAVE;
        if (e < a.d) a.params[(long)node * a.d + e] = w[r];
    }
    if (lane == 0) a.ages[node] = age;
}

__global__ void __launch_bounds__(WAVE)
tick_linear_kernel(LinearArgs a)
{
    linear_process_node(a, blockIdx.x);
}

// ---------------------------------------------------------------------------
// MLP tick: one workgroup per node, all activations in LDS
// ---------------------------------------------------------------------------

struct MlpArgs {
    float* params; int* ages;
    float* slots; int* slot_ages;
    const int* nodes; const int* ptr;
    const int* dslots; const int* rslots;
    const int* dmodes;
    const float* X; const float* Y; const int* counts;
    const int* layout;  // [n_layers][4] = (w_off, b_off, in, out)
    int n_layers, Smax, D, act_max, d_in;
    float lr, wd;
    int epochs, bs, mode, update_only;
};

// Fused fwd+bwd+SGD over the node's shard. LDS holds the model row plus
// per-batch activations of every layer (act) and their gradients (grad).
//
// All three GEMMs (forward X·Wᵀ, input-grad G·W, weight-grad Gᵀ·X) run on
// the matrix cores via v_mfma_f32_16x16x4_f32 (exact f32 — bitwise an fmaf
// chain, so the torch oracle tolerance is unchanged). Fragment maps per
// the CDNA4 ISA: lane l supplies A[l&15][l>>4] and B[l>>4][l&15]; the
// accumulator lane l, register r holds D[(l>>4)*4 + r][l&15]. Ragged tile
// edges are zero-padded in registers (exact for f32). One 16x16 output
// tile per wave per iteration; the block's 4 waves stride the tile grid.

typedef float f32x4_t __attribute__((ext_vector_type(4)));

// Dual-accumulator MFMA K-loop. For K <= MLP_KPAD the loop is PADDED to a
// compile-time bound and fully unrolled: every LDS load issues up front
// (independent) and the two MFMA chains run issue-bound instead of
// load-latency-bound. Out-of-range chunks contribute exact zeros, and the
// chunk->accumulator pairing (k0%8==0 -> acc, ==4 -> acc2) matches the
// dynamic loop, so the result is BIT-IDENTICAL either way. LoadA/LoadB
// take the chunk base k0 and must return 0 outside [0, Kdim).
constexpr int MLP_KPAD = 128;

template <int PAD, typename LoadA, typename LoadB>
DEV_INLINE f32x4_t mfma_kloop_padded(LoadA la, LoadB lb)
{
    f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
    f32x4_t acc2 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int k0 = 0; k0 < PAD; k0 += 8) {
        float av = la(k0), bv = lb(k0);
        float av2 = la(k0 + 4), bv2 = lb(k0 + 4);
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(av, bv, acc, 0, 0, 0);
        acc2 = __builtin_amdgcn_mfma_f32_16x16x4f32(av2, bv2, acc2, 0, 0, 0);
    }
    for (int r = 0; r < 4; ++r) acc[r] += acc2[r];
    return acc;
}

template <typename LoadA, typename LoadB>
DEV_INLINE f32x4_t mfma_kloop(int Kdim, LoadA la, LoadB lb)
{
    if (Kdim <= 32) return mfma_kloop_padded<32>(la, lb);
    if (Kdim <= 64) return mfma_kloop_padded<64>(la, lb);
    if (Kdim <= MLP_KPAD) return mfma_kloop_padded<MLP_KPAD>(la, lb);
    f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
    f32x4_t acc2 = {0.f, 0.f, 0.f, 0.f};
    {
        int k0 = 0;
        for (; k0 + 8 <= Kdim; k0 += 8) {
            float av = la(k0), bv = lb(k0);
            float av2 = la(k0 + 4), bv2 = lb(k0 + 4);
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(av, bv, acc, 0, 0, 0);
            acc2 = __builtin_amdgcn_mfma_f32_16x16x4f32(
                av2, bv2, acc2, 0, 0, 0);
        }
        for (; k0 < Kdim; k0 += 4) {
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(
                la(k0), lb(k0), acc, 0, 0, 0);
        }
    }
    for (int r = 0; r < 4; ++r) acc[r] += acc2[r];
    return acc;
}

DEV_INLINE void mlp_update(const MlpArgs& a, int node, float* W,
                           float* act, float* grad, int& age)
{
    int tid = threadIdx.x;
    int c = a.counts[node];
    if (c == 0) return;
    int bsz = (a.bs == 0) ? c : min(a.bs, c);
    const float* Xn = a.X + (long)node * a.Smax * a.d_in;
    const float* Yn = a.Y + (long)node * a.Smax;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int fr = lane & 15;       // A-row / B-col fragment index
    const int fk = lane >> 4;       // K fragment index (0..3)
    const int nwaves = blockDim.x >> 6;
    for (int ep = 0; ep < a.epochs; ++ep) {
        for (int s0 = 0; s0 < c; s0 += bsz) {
            int m = min(bsz, c - s0);
            // activations laid out layer after layer: act[l] is [m, out_l];
            // act[-1] (the input) is staged first at act.
            float* in = act;
            for (int e = tid; e < m * a.d_in; e += blockDim.x)
                in[e] = Xn[(long)s0 * a.d_in + e];
            __syncthreads();
            // ---- forward: H = X · Wᵀ + b on MFMA
            float* cur = in;
            float* nxt = act + m * a.d_in;
            for (int l = 0; l < a.n_layers; ++l) {
                int w_off = a.layout[4 * l], b_off = a.layout[4 * l + 1];
                int fin = a.layout[4 * l + 2], fout = a.layout[4 * l + 3];
                int m16 = (m + 15) >> 4, o16 = (fout + 15) >> 4;
                for (int tile = wave; tile < m16 * o16; tile += nwaves) {
                    int tr = (tile / o16) << 4, tc = (tile % o16) << 4;
                    f32x4_t acc = mfma_kloop(
                        fin,
                        [&](int k0) {
                            int q = k0 + fk;
                            return (tr + fr < m && q < fin)
                                       ? cur[(tr + fr) * fin + q] : 0.f;
                        },
                        [&](int k0) {
                            int q = k0 + fk;
                            return (tc + fr < fout && q < fin)
                                       ? W[w_off + (tc + fr) * fin + q]
                                       : 0.f;
                        });
                    for (int r = 0; r < 4; ++r) {
                        int row = tr + (fk << 2) + r, col = tc + fr;
                        if (row < m && col < fout) {
                            float v = acc[r] + W[b_off + col];
                            if (l < a.n_layers - 1) v = fmaxf(v, 0.f);  // ReLU
                            nxt[row * fout + col] = v;
                        }
                    }
                }
                __syncthreads();
                cur = nxt;
                nxt = nxt + m * fout;
            }
            // ---- output grad: softmax-CE on the raw head
            int k = a.layout[4 * (a.n_layers - 1) + 3];
            float* gcur = grad;  // [m, k] for the head, reused per layer
            if (tid < m) {
                const float* z = cur + tid * k;
                float zmax = -1e30f;
                for (int kk = 0; kk < k; ++kk) zmax = fmaxf(zmax, z[kk]);
                float sum = 0.f;
                for (int kk = 0; kk < k; ++kk) sum += __expf(z[kk] - zmax);
                int yi = (int)Yn[s0 + tid];
                float inv = 1.0f / sum;
                for (int kk = 0; kk < k; ++kk) {
                    float p = __expf(z[kk] - zmax) * inv;
                    gcur[tid * k + kk] = (p - (kk == yi ? 1.0f : 0.0f)) / m;
                }
            }
            __syncthreads();
            // ---- backward + SGD, layer by layer (grad buffers ping-pong);
            //      both GEMMs (dX = G·W, dW = Gᵀ·X) on MFMA
            float* gnext = grad + m * a.act_max;
            for (int l = a.n_layers - 1; l >= 0; --l) {
                int w_off = a.layout[4 * l], b_off = a.layout[4 * l + 1];
                int fin = a.layout[4 * l + 2], fout = a.layout[4 * l + 3];
                // activation input of this layer
                float* ain = act;
                for (int q = 0; q < l; ++q) ain += m * a.layout[4 * q + 2];
                // (ain now points at act of layer l's input: layers are
                //  packed input-first, so offset = m*(d_in + hidden_0 + ...))
                // grad wrt input (needed before W is updated):
                // dX[m, fin] = G[m, fout] · W[fout, fin], ReLU-masked
                if (l > 0) {
                    int m16 = (m + 15) >> 4, q16 = (fin + 15) >> 4;
                    for (int tile = wave; tile < m16 * q16; tile += nwaves) {
                        int tr = (tile / q16) << 4, tc = (tile % q16) << 4;
                        f32x4_t acc = mfma_kloop(
                            fout,
                            [&](int k0) {
                                int o = k0 + fk;
                                return (tr + fr < m && o < fout)
                                           ? gcur[(tr + fr) * fout + o]
                                           : 0.f;
                            },
                            [&](int k0) {
                                int o = k0 + fk;
                                return (o < fout && tc + fr < fin)
                                           ? W[w_off + o * fin + tc + fr]
                                           : 0.f;
                            });
                        for (int r = 0; r < 4; ++r) {
                            int row = tr + (fk << 2) + r, col = tc + fr;
                            if (row < m && col < fin) {
                                float v = acc[r];
                                // ReLU mask of the layer-(l-1) activation
                                v *= (ain[row * fin + col] > 0.f) ? 1.0f : 0.0f;
                                gnext[row * fin + col] = v;
                            }
                        }
                    }
                    __syncthreads();
                }
                // weight SGD: dW[fout, fin] = Gᵀ[fout, m] · X[m, fin]
                {
                    int o16 = (fout + 15) >> 4, q16 = (fin + 15) >> 4;
                    for (int tile = wave; tile < o16 * q16; tile += nwaves) {
                        int tr = (tile / q16) << 4, tc = (tile % q16) << 4;
                        f32x4_t acc = mfma_kloop(
                            m,
                            [&](int k0) {
                                int sidx = k0 + fk;
                                return (tr + fr < fout && sidx < m)
                                           ? gcur[sidx * fout + tr + fr]
                                           : 0.f;
                            },
                            [&](int k0) {
                                int sidx = k0 + fk;
                                return (sidx < m && tc + fr < fin)
                                           ? ain[sidx * fin + tc + fr]
                                           : 0.f;
                            });
                        for (int r = 0; r < 4; ++r) {
                            int row = tr + (fk << 2) + r, col = tc + fr;
                            if (row < fout && col < fin) {
                                float g = acc[r];
                                int e = row * fin + col;
                                if (a.wd != 0.f) g += a.wd * W[w_off + e];
                                W[w_off + e] -= a.lr * g;
                            }
                        }
                    }
                }
                for (int e = tid; e < fout; e += blockDim.x) {
                    float g = 0.f;
                    for (int s = 0; s < m; ++s) g += gcur[s * fout + e];
                    W[b_off + e] -= a.lr * g;
                }
                __syncthreads();
                float* tmp = gcur; gcur = gnext; gnext = tmp;
            }
            age += 1;
        }
    }
}

__global__ void __launch_bounds__(512)
tick_mlp_kernel(MlpArgs a)
{
    int i = blockIdx.x;
    int node = a.nodes[i];
    int tid = threadIdx.x;
    extern __shared__ float sm[];
    float* W = sm;          // D
    // W2 (the second model slab) exists only for UPDATE_MERGE — dropping
    // it elsewhere takes the flagship MLP block from 94 KB to 70 KB of
    // LDS, which fits TWO workgroups per CU (4 waves/SIMD instead of 2)
    bool has_w2 = (a.mode == MODE_UPDATE_MERGE);
    float* W2 = has_w2 ? W + a.D : nullptr;
    int bsmax = (a.bs == 0) ? a.Smax : min(a.bs, a.Smax);
    // act: batch input + every layer's activation; grad: 2 ping-pong buffers
    float* act = W + (has_w2 ? 2 : 1) * (size_t)a.D;
    int act_total = 0;
    // act layout computed on host side == bs*(d_in + sum(out_l)); the host
    // passes act_max = max layer width for the grad buffers
    for (int l = 0; l < a.n_layers; ++l) act_total += a.layout[4 * l + 3];
    float* grad = act + bsmax * (a.d_in + act_total);

    for (int e = tid; e < a.D; e += blockDim.x)
        W[e] = a.params[(long)node * a.D + e];
    __syncthreads();
    int age = a.ages[node];

    if (a.update_only) {
        mlp_update(a, node, W, act, grad, age);
    } else {
        for (int j = a.ptr[i]; j < a.ptr[i + 1]; ++j) {
            int slot = a.dslots[j];
            const float* srow = a.slots + (long)slot * a.D;
            int sage = a.slot_ages[slot];
            int md = (a.dmodes && a.dmodes[j]) ? MODE_PASS : a.mode;
            if (md == MODE_MERGE_UPDATE) {
                for (int e = tid; e < a.D; e += blockDim.x)
                    W[e] = 0.5f * (W[e] + srow[e]);
                age = max(age, sage);
                __syncthreads();
                mlp_update(a, node, W, act, grad, age);
            } else if (md == MODE_UPDATE) {
                for (int e = tid; e < a.D; e += blockDim.x) W[e] = srow[e];
                age = sage;
                __syncthreads();
                mlp_update(a, node, W, act, grad, age);
            } else if (md == MODE_UPDATE_MERGE) {
                mlp_update(a, node, W, act, grad, age);
                for (int e = tid; e < a.D; e += blockDim.x) W2[e] = srow[e];
                __syncthreads();
                int age2 = sage;
                mlp_update(a, node, W2, act, grad, age2);
                for (int e = tid; e < a.D; e += blockDim.x)
                    W[e] = 0.5f * (W[e] + W2[e]);
                age = max(age, age2);
                __syncthreads();
            } else {
                for (int e = tid; e < a.D; e += blockDim.x) W[e] = srow[e];
                age = sage;
                __syncthreads();
            }
            int rs = a.rslots ? a.rslots[j] : -1;
            if (rs >= 0) {
                for (int e = tid; e < a.D; e += blockDim.x)
                    a.slots[(long)rs * a.D + e] = W[e];
                if (tid == 0) a.slot_ages[rs] = age;
                __syncthreads();
            }
        }
    }
    __syncthreads();
    for (int e = tid; e < a.D; e += blockDim.x)
        a.params[(long)node * a.D + e] = W[e];
    if (tid == 0) a.ages[node] = age;
}

// ---------------------------------------------------------------------------
// fused evaluation metrics (K13): accuracy / macro precision / recall / F1
// (+ pairwise ROC-AUC for binary) of R node models on one shared eval set,
// in a single launch. Replaces the reference's per-node sklearn calls
// (gossipy/model/handler.py:282-334, 375-391) and the engine's previous
// ~15-small-torch-ops path. argmax/AUC are computed on raw affine scores —
// the sigmoid is monotonic, so predictions and ranks are unchanged.
// ---------------------------------------------------------------------------

struct EvalArgs {
    const float* params;  // [n_local, D]
    const int* nodes;     // [R] local row ids
    const float* X;       // [n_eval, d]
    const float* Y;       // [n_eval] class index (or +-1 when is_margin)
    float* out;           // [R, 5] acc, prec, rec, f1, auc
    int d, k, n_eval, D;
    int is_margin;        // pegasos/adaline: score = <w, x>, label +-1
    int tile;             // samples staged per LDS tile (coalesced loads)
    // per-node shard mode (local test sets): X/Y are [n_local, n_eval(,d)]
    // arenas, eval_counts[node] gives the node's row count
    const int* eval_counts;
    // precomputed-scores mode (MLP/torchmod: any family whose forward ran
    // elsewhere): [R, n_eval, k] raw scores; skips the scoring GEMM stage
    const float* scores;
};

constexpr int EVAL_KMAX = 16;

__global__ void __launch_bounds__(256)
eval_metrics_kernel(EvalArgs a)
{
    int r = blockIdx.x;
    int node = a.nodes ? a.nodes[r] : r;
    int tid = threadIdx.x;
    int n_e = a.n_eval;
    const float* Xb = a.X;
    const float* Yb = a.Y;
    if (a.eval_counts) {  // per-node shard mode: a.n_eval is the row stride
        Xb = a.X + (long)node * a.n_eval * a.d;
        Yb = a.Y + (long)node * a.n_eval;
        n_e = min(a.eval_counts[node], a.n_eval);
        if (n_e == 0) {
            if (tid == 0)
                for (int e = 0; e < 5; ++e) a.out[(long)r * 5 + e] = -2.f;
            return;
        }
    }
    extern __shared__ float sm[];
    float* W = sm;                       // D
    float* s1 = W + a.D;                 // n_eval: positive-class score
    char* yb = (char*)(s1 + n_e);   // n_eval: binarized label
    int* conf = (int*)(yb + ((n_e + 15) & ~15));  // k*k confusion
    float* XT = (float*)(conf + EVAL_KMAX * EVAL_KMAX);  // tile x d
    int kk = a.is_margin ? 2 : a.k;

    for (int e = tid; e < a.D; e += blockDim.x)
        W[e] = a.params[(long)node * a.D + e];
    for (int e = tid; e < kk * kk; e += blockDim.x) conf[e] = 0;
    __syncthreads();

    if (a.scores) {
        // precomputed scores: just argmax + confusion + score staging
        const float* zs = a.scores + (long)r * n_e * (a.is_margin ? 1 : a.k);
        for (int sidx = tid; sidx < n_e; sidx += blockDim.x) {
            int pred, yt;
            float sc1;
            if (a.is_margin) {
                sc1 = zs[sidx];
                pred = sc1 >= 0.f ? 1 : 0;
                yt = Yb[sidx] > 0.f ? 1 : 0;
            } else {
                const float* z = zs + (long)sidx * a.k;
                float best = -1e30f;
                int bj = 0;
                for (int j = 0; j < a.k; ++j)
                    if (z[j] > best) { best = z[j]; bj = j; }
                sc1 = a.k > 1 ? z[1] : z[0];
                pred = bj;
                yt = (int)Yb[sidx];
            }
            s1[sidx] = sc1;
            yb[sidx] = (char)(a.is_margin ? (Yb[sidx] > 0.f ? 1 : 0)
                                          : ((int)Yb[sidx] == 1 ? 1 : 0));
            atomicAdd(&conf[yt * kk + pred], 1);
        }
        __syncthreads();
    } else
    // samples staged through LDS in coalesced tiles (the old thread-per-
    // sample global gather was a stride-d access pattern and dominated
    // the kernel's 44 us)
    for (int s0 = 0; s0 < n_e; s0 += a.tile) {
        int m = min(a.tile, n_e - s0);
        for (int e = tid; e < m * a.d; e += blockDim.x)
            XT[e] = Xb[(long)s0 * a.d + e];
        __syncthreads();
        for (int sidx = s0 + tid; sidx < s0 + m; sidx += blockDim.x) {
            const float* x = XT + (long)(sidx - s0) * a.d;
            int pred, yt;
            float sc1;
            if (a.is_margin) {
                float acc = 0.f;
                for (int e = 0; e < a.d; ++e) acc += W[e] * x[e];
                sc1 = acc;
                pred = acc >= 0.f ? 1 : 0;
                yt = Yb[sidx] > 0.f ? 1 : 0;
            } else {
                float best = -1e30f;
                int bj = 0;
                float z1 = 0.f;
                for (int j = 0; j < a.k; ++j) {
                    float acc = W[a.k * a.d + j];
                    const float* wrow = W + j * a.d;
                    for (int e = 0; e < a.d; ++e) acc += wrow[e] * x[e];
                    if (acc > best) { best = acc; bj = j; }
                    if (j == 1) z1 = acc;
                }
                sc1 = z1;
                pred = bj;
                yt = (int)Yb[sidx];
            }
            s1[sidx] = sc1;
            yb[sidx] = (char)(a.is_margin ? (Yb[sidx] > 0.f ? 1 : 0)
                                          : ((int)Yb[sidx] == 1 ? 1 : 0));
            atomicAdd(&conf[yt * kk + pred], 1);
        }
        __syncthreads();
    }

    // pairwise AUC (binary only): wins over (pos, neg) pairs, ties 0.5 —
    // the Mann-Whitney statistic (== average-rank AUC)
    __shared__ float auc_wins;
    __shared__ int npos_s;
    if (tid == 0) { auc_wins = 0.f; npos_s = 0; }
    __syncthreads();
    if (kk == 2) {
        float wins = 0.f;
        int npos_l = 0;
        for (int i = tid; i < n_e; i += blockDim.x) {
            if (!yb[i]) continue;
            npos_l += 1;
            for (int j = 0; j < n_e; ++j) {
                if (yb[j]) continue;
                if (s1[i] > s1[j]) wins += 1.f;
                else if (s1[i] == s1[j]) wins += 0.5f;
            }
        }
        // block reduce via wave shuffles: 256 serialized LDS atomics were
        // ~1/3 of this kernel's time
        wins = wave_sum(wins);
        float np_f = wave_sum((float)npos_l);
        if ((tid & 63) == 0) {
            atomicAdd(&auc_wins, wins);
            atomicAdd(&npos_s, (int)np_f);
        }
    }
    __syncthreads();

    if (tid == 0) {
        int n = n_e;
        int correct = 0;
        float prec = 0.f, rec = 0.f, f1 = 0.f;
        for (int c = 0; c < kk; ++c) {
            int tp = conf[c * kk + c];
            correct += tp;
            int pred_tot = 0, true_tot = 0;
            for (int t2 = 0; t2 < kk; ++t2) {
                pred_tot += conf[t2 * kk + c];
                true_tot += conf[c * kk + t2];
            }
            float p = pred_tot > 0 ? (float)tp / pred_tot : 0.f;
            float q = true_tot > 0 ? (float)tp / true_tot : 0.f;
            prec += p;
            rec += q;
            f1 += (p + q) > 0.f ? 2.f * p * q / (p + q) : 0.f;
        }
        float* o = a.out + (long)r * 5;
        o[0] = (float)correct / n;
        o[1] = prec / kk;
        o[2] = rec / kk;
        o[3] = f1 / kk;
        if (kk == 2) {
            int npos = npos_s, nneg = n - npos_s;
            o[4] = (npos == 0 || nneg == 0)
                       ? 0.5f
                       : auc_wins / ((float)npos * nneg);
        } else {
            o[4] = -1.f;  // AUC undefined for k > 2 (reference parity)
        }
    }
}

// ---------------------------------------------------------------------------
// PENS step-1 event kernel (Onoszko 2021; gossipy/node.py:767-782): one
// workgroup per event — score all cached candidate models on the
// receiver's train shard (argmax accuracy; the sigmoid is monotonic so
// raw affine scores give the reference's predictions), select the top-m
// (stable in arrival order), merge them as a (m+1)-way mean with the own
// model, run the local update, and count the winning senders.
// ---------------------------------------------------------------------------

constexpr int PENS_MAX_CAND = 64;

struct PensArgs {
    float* params; int* ages;
    const float* slots; const int* slot_ages;
    const int* nodes;       // [E] event receivers (unique within a launch)
    const int* ptr;         // [E+1] candidate ranges
    const int* cslots;      // candidate slots
    const int* cowners;     // candidate senders (global node ids)
    int* counts;            // [n_local, n_total] winner counters
    const float* X; const float* Y; const int* dcounts;
    int d, k, Smax, D, m_top, n_total;
    float lr, wd;
    int epochs, bs;
};

__global__ void __launch_bounds__(128)
tick_pens_kernel(PensArgs a)
{
    int i = blockIdx.x;
    int node = a.nodes[i];
    int tid = threadIdx.x;
    int lo = a.ptr[i], hi = a.ptr[i + 1];
    int n_cand = hi - lo;
    extern __shared__ float sm[];
    float* W = sm;                      // D
    float* xb = W + a.D;                // bsmax*d (update scratch)
    int bsmax = (a.bs == 0) ? a.Smax : min(a.bs, a.Smax);
    float* dz = xb + bsmax * a.d;       // bsmax*k
    __shared__ int correct[PENS_MAX_CAND];
    __shared__ int sel[PENS_MAX_CAND];

    int c_n = a.dcounts[node];
    const float* Xn = a.X + (long)node * a.Smax * a.d;
    const float* Yn = a.Y + (long)node * a.Smax;

    // --- score candidates: thread = sample, loop candidates
    for (int c = tid; c < n_cand; c += blockDim.x) correct[c] = 0;
    __syncthreads();
    for (int c = 0; c < n_cand; ++c) {
        const float* Wc = a.slots + (long)a.cslots[lo + c] * a.D;
        int my = 0;
        for (int s = tid; s < c_n; s += blockDim.x) {
            const float* x = Xn + (long)s * a.d;
            float best = -1e30f;
            int bj = 0;
            for (int j = 0; j < a.k; ++j) {
                float acc = Wc[a.k * a.d + j];
                const float* wrow = Wc + j * a.d;
                for (int e = 0; e < a.d; ++e) acc += wrow[e] * x[e];
                if (acc > best) { best = acc; bj = j; }
            }
            if (bj == (int)Yn[s]) my += 1;
        }
        if (my) atomicAdd(&correct[c], my);
    }
    __syncthreads();

    // --- stable top-m selection (thread 0; n_cand small)
    if (tid == 0) {
        for (int c = 0; c < n_cand; ++c) sel[c] = 0;
        int m = min(a.m_top, n_cand);
        for (int pick = 0; pick < m; ++pick) {
            int best_c = -1, best_v = -1;
            for (int c = 0; c < n_cand; ++c) {
                if (!sel[c] && correct[c] > best_v) {
                    best_v = correct[c];
                    best_c = c;
                }
            }
            sel[best_c] = 1;
            atomicAdd(&a.counts[(long)node * a.n_total + a.cowners[lo + best_c]], 1);
        }
    }
    __syncthreads();

    // --- merge: (own + sum selected) / (m+1), age = max
    for (int e = tid; e < a.D; e += blockDim.x)
        W[e] = a.params[(long)node * a.D + e];
    __syncthreads();
    int n_sel = 0;
    int age = a.ages[node];
    for (int c = 0; c < n_cand; ++c) {
        if (!sel[c]) continue;
        n_sel += 1;
        const float* Wc = a.slots + (long)a.cslots[lo + c] * a.D;
        for (int e = tid; e < a.D; e += blockDim.x) W[e] += Wc[e];
        age = max(age, a.slot_ages[a.cslots[lo + c]]);
    }
    float inv = 1.0f / (n_sel + 1);
    for (int e = tid; e < a.D; e += blockDim.x) W[e] *= inv;
    __syncthreads();

    // --- local update
    LogregArgs u;
    u.X = a.X; u.Y = a.Y; u.counts = a.dcounts;
    u.d = a.d; u.k = a.k; u.Smax = a.Smax; u.D = a.D;
    u.lr = a.lr; u.wd = a.wd; u.epochs = a.epochs; u.bs = a.bs;
    logreg_update(u, node, W, xb, dz, age);
    __syncthreads();
    for (int e = tid; e < a.D; e += blockDim.x)
        a.params[(long)node * a.D + e] = W[e];
    if (tid == 0) a.ages[node] = age;
}

// ---------------------------------------------------------------------------
// cooperative whole-round kernel (logreg family): the entire round — every
// tick's snapshot and delivery batch — runs inside ONE kernel launch, with
// grid.sync() as the tick barrier. Grids here are tiny (max batch ≈ a few
// dozen blocks on 256 CUs), so cooperative residency is guaranteed and the
// launch is REJECTED (not deadlocked) if ever oversubscribed. Cuts the
// ~200 per-round launch overheads of the stream executor to one.
// ---------------------------------------------------------------------------

struct CoopRoundArgs {
    LogregArgs base;  // nodes/ptr/dslots/rslots are set per group inside
    const int* snap_nodes; const int* snap_slots; const int* snap_tptr;
    const int* recv_nodes; const int* recv_nptr; const int* recv_tptr;
    const int* del_slots; const int* reply_slots;
    const int* pull_nodes; const int* pull_slots; const int* pull_tptr;
    const int* rep_nodes; const int* rep_nptr; const int* rep_tptr;
    const int* rep_slots;
    int delta;
};

// grid barrier that degrades to a block barrier when the kernel was
// launched PLAIN with one workgroup (the single-block round path for
// tiny-batch schedules — one launch per round, no cooperative API)
DEV_INLINE void round_sync()
{
    if (gridDim.x == 1) {
        __syncthreads();
    } else {
        cooperative_groups::this_grid().sync();
    }
}

__global__ void __launch_bounds__(128)
coop_round_logreg_kernel(CoopRoundArgs c)
{
    LogregArgs a = c.base;
    const int nb = gridDim.x;
    const int bid = blockIdx.x;
    const int tid = threadIdx.x;
    bool dirty = false;  // writes since the last grid sync

    for (int t = 0; t < c.delta; ++t) {
        int s0 = c.snap_tptr[t], s1 = c.snap_tptr[t + 1];
        int r0 = c.recv_tptr[t], r1 = c.recv_tptr[t + 1];
        int p0 = c.pull_tptr[t], p1 = c.pull_tptr[t + 1];
        int q0 = c.rep_tptr[t], q1 = c.rep_tptr[t + 1];

        if (s1 > s0) {
            if (dirty) { round_sync(); dirty = false; }
            // block-strided row copies (the snapshot sub-phase)
            for (int i = s0 + bid; i < s1; i += nb) {
                int node = c.snap_nodes[i];
                int slot = c.snap_slots[i];
                for (int e = tid; e < a.D; e += blockDim.x)
                    a.slots[(long)slot * a.D + e] =
                        a.params[(long)node * a.D + e];
                if (tid == 0) a.slot_ages[slot] = a.ages[node];
            }
            dirty = true;
        }
        if (r1 > r0) {
            if (dirty) { round_sync(); dirty = false; }
            a.nodes = c.recv_nodes;
            a.ptr = c.recv_nptr;
            a.dslots = c.del_slots;
            a.rslots = c.reply_slots;
            for (int i = r0 + bid; i < r1; i += nb) {
                logreg_process_node(a, i);
                __syncthreads();
            }
            dirty = true;
        }
        if (p1 > p0) {
            if (dirty) { round_sync(); dirty = false; }
            for (int i = p0 + bid; i < p1; i += nb) {
                int node = c.pull_nodes[i];
                int slot = c.pull_slots[i];
                for (int e = tid; e < a.D; e += blockDim.x)
                    a.slots[(long)slot * a.D + e] =
                        a.params[(long)node * a.D + e];
                if (tid == 0) a.slot_ages[slot] = a.ages[node];
            }
            dirty = true;
        }
        if (q1 > q0) {
            if (dirty) { round_sync(); dirty = false; }
            a.nodes = c.rep_nodes;
            a.ptr = c.rep_nptr;
            a.dslots = c.rep_slots;
            a.rslots = nullptr;
            for (int i = q0 + bid; i < q1; i += nb) {
                logreg_process_node(a, i);
                __syncthreads();
            }
            dirty = true;
        }
    }
}

// ---------------------------------------------------------------------------
// single-block round kernels (linear / partitioned): for schedules whose
// per-tick batches are tiny (async gossip, tokenized 100-node configs),
// the whole round runs as ONE plain launch of ONE workgroup —
// __syncthreads() is the tick barrier, replacing ~2 launches per tick.
// ---------------------------------------------------------------------------

struct LinRoundArgs {
    LinearArgs base;
    const int* snap_nodes; const int* snap_slots; const int* snap_tptr;
    const int* recv_nodes; const int* recv_nptr; const int* recv_tptr;
    const int* del_slots; const int* reply_slots;
    const int* pull_nodes; const int* pull_slots; const int* pull_tptr;
    const int* rep_nodes; const int* rep_nptr; const int* rep_tptr;
    const int* rep_slots;
    int delta;
};

__global__ void __launch_bounds__(WAVE)
sb_round_linear_kernel(LinRoundArgs c)
{
    LinearArgs a = c.base;
    int tid = threadIdx.x;
    for (int t = 0; t < c.delta; ++t) {
        int s0 = c.snap_tptr[t], s1 = c.snap_tptr[t + 1];
        for (int i = s0; i < s1; ++i) {
            int node = c.snap_nodes[i], slot = c.snap_slots[i];
            for (int e = tid; e < a.d; e += blockDim.x)
                a.slots[(long)slot * a.d + e] = a.params[(long)node * a.d + e];
            if (tid == 0) a.slot_ages[slot] = a.ages[node];
        }
        __syncthreads();
        int r0 = c.recv_tptr[t], r1 = c.recv_tptr[t + 1];
        if (r1 > r0) {
            a.nodes = c.recv_nodes;
            a.ptr = c.recv_nptr;
            a.dslots = c.del_slots;
            a.rslots = c.reply_slots;
            for (int i = r0; i < r1; ++i) linear_process_node(a, i);
            __syncthreads();
        }
        int p0 = c.pull_tptr[t], p1 = c.pull_tptr[t + 1];
        for (int i = p0; i < p1; ++i) {
            int node = c.pull_nodes[i], slot = c.pull_slots[i];
            for (int e = tid; e < a.d; e += blockDim.x)
                a.slots[(long)slot * a.d + e] = a.params[(long)node * a.d + e];
            if (tid == 0) a.slot_ages[slot] = a.ages[node];
        }
        if (p1 > p0) __syncthreads();
        int q0 = c.rep_tptr[t], q1 = c.rep_tptr[t + 1];
        if (q1 > q0) {
            a.nodes = c.rep_nodes;
            a.ptr = c.rep_nptr;
            a.dslots = c.rep_slots;
            a.rslots = nullptr;
            for (int i = q0; i < q1; ++i) linear_process_node(a, i);
            __syncthreads();
        }
    }
}

struct PartRoundArgs {
    LogregPartArgs base;
    const int* snap_nodes; const int* snap_slots; const int* snap_tptr;
    const int* recv_nodes; const int* recv_nptr; const int* recv_tptr;
    const int* del_slots; const int* reply_slots; const int* del_pids;
    const int* pull_nodes; const int* pull_slots; const int* pull_tptr;
    const int* rep_nodes; const int* rep_nptr; const int* rep_tptr;
    const int* rep_slots; const int* rep_pids;
    int delta;
};

__global__ void __launch_bounds__(128)
sb_round_logreg_part_kernel(PartRoundArgs c)
{
    LogregPartArgs a = c.base;
    int tid = threadIdx.x;
    for (int t = 0; t < c.delta; ++t) {
        int s0 = c.snap_tptr[t], s1 = c.snap_tptr[t + 1];
        for (int i = s0; i < s1; ++i) {
            int node = c.snap_nodes[i], slot = c.snap_slots[i];
            for (int e = tid; e < a.D; e += blockDim.x)
                a.slots[(long)slot * a.D + e] = a.params[(long)node * a.D + e];
            for (int p = tid; p < a.P; p += blockDim.x)
                a.slot_ages[(long)slot * a.P + p] = a.ages[(long)node * a.P + p];
        }
        __syncthreads();
        int r0 = c.recv_tptr[t], r1 = c.recv_tptr[t + 1];
        if (r1 > r0) {
            a.nodes = c.recv_nodes;
            a.ptr = c.recv_nptr;
            a.dslots = c.del_slots;
            a.rslots = c.reply_slots;
            a.dpids = c.del_pids;
            for (int i = r0; i < r1; ++i) {
                logreg_part_process_node(a, i);
                __syncthreads();
            }
        }
        int p0 = c.pull_tptr[t], p1 = c.pull_tptr[t + 1];
        for (int i = p0; i < p1; ++i) {
            int node = c.pull_nodes[i], slot = c.pull_slots[i];
            for (int e = tid; e < a.D; e += blockDim.x)
                a.slots[(long)slot * a.D + e] = a.params[(long)node * a.D + e];
            for (int p = tid; p < a.P; p += blockDim.x)
                a.slot_ages[(long)slot * a.P + p] = a.ages[(long)node * a.P + p];
        }
        if (p1 > p0) __syncthreads();
        int q0 = c.rep_tptr[t], q1 = c.rep_tptr[t + 1];
        if (q1 > q0) {
            a.nodes = c.rep_nodes;
            a.ptr = c.rep_nptr;
            a.dslots = c.rep_slots;
            a.rslots = nullptr;
            a.dpids = c.rep_pids;
            for (int i = q0; i < q1; ++i) {
                logreg_part_process_node(a, i);
                __syncthreads();
            }
        }
    }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

#define CHECK_DEV(t) TORCH_CHECK(t.is_cuda() && t.is_contiguous(), #t " must be contiguous on device")

static hipStream_t current_stream()
{
    return at::hip::getCurrentHIPStream().stream();
}

void snapshot(torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
              torch::Tensor slot_ages, torch::Tensor nodes, torch::Tensor slot_ids,
              int64_t age_width, int64_t src_off)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(nodes); CHECK_DEV(slot_ids);
    int n = nodes.size(0);
    if (n == 0) return;
    int W = slots.size(1);
    int Dp = params.size(1);
    long total = (long)n * W;
    int block = 256;
    int grid = (int)std::min<long>((total + block - 1) / block, 2048);
    hipLaunchKernelGGL(snapshot_kernel, dim3(grid), dim3(block), 0, current_stream(),
        params.data_ptr<float>(), ages.data_ptr<int>(),
        slots.data_ptr<float>(), slot_ages.data_ptr<int>(),
        nodes.data_ptr<int>(), slot_ids.data_ptr<int>(), n, W, Dp,
        (int)src_off, (int)age_width);
}

void wmerge(torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
            torch::Tensor slot_ages, torch::Tensor nodes, torch::Tensor ptr,
            torch::Tensor wslots, torch::Tensor wweights, torch::Tensor selfw)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(nodes); CHECK_DEV(ptr);
    int n = nodes.size(0);
    if (n == 0) return;
    int D = params.size(1);
    hipLaunchKernelGGL(wmerge_kernel, dim3(n), dim3(128), 0, current_stream(),
        params.data_ptr<float>(), ages.data_ptr<int>(),
        slots.data_ptr<float>(), slot_ages.data_ptr<int>(),
        nodes.data_ptr<int>(), ptr.data_ptr<int>(),
        wslots.numel() ? wslots.data_ptr<int>() : nullptr,
        wweights.numel() ? wweights.data_ptr<float>() : nullptr,
        selfw.data_ptr<float>(), D);
}

void tick_logreg_part(
    torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
    torch::Tensor slot_ages, torch::Tensor nodes, torch::Tensor recv_ptr,
    torch::Tensor del_slots, torch::Tensor reply_slots, torch::Tensor del_pids,
    torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
    torch::Tensor perm, torch::Tensor pptr, torch::Tensor apart,
    int64_t n_parts, int64_t d, int64_t k, double lr, double wd,
    int64_t epochs, int64_t bs, int64_t mode, bool update_only)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X); CHECK_DEV(Y);
    CHECK_DEV(perm); CHECK_DEV(pptr); CHECK_DEV(apart);
    TORCH_CHECK(k <= KMAX, "n_classes > ", KMAX, " unsupported");
    TORCH_CHECK(mode != MODE_PASS, "Mode PASS not allowed for partitioned models.");
    int n = nodes.size(0);
    if (n == 0) return;
    LogregPartArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.nodes = nodes.data_ptr<int>(); a.ptr = recv_ptr.data_ptr<int>();
    a.dslots = del_slots.numel() ? del_slots.data_ptr<int>() : nullptr;
    a.rslots = reply_slots.numel() ? reply_slots.data_ptr<int>() : nullptr;
    a.dpids = del_pids.numel() ? del_pids.data_ptr<int>() : nullptr;
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.perm = perm.data_ptr<int>(); a.pptr = pptr.data_ptr<int>();
    a.apart = apart.data_ptr<int>();
    a.P = n_parts; a.d = d; a.k = k; a.Smax = X.size(1); a.D = params.size(1);
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs; a.mode = mode;
    a.update_only = update_only;
    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    // whole-shard staging when it fits a 64 KB block budget (keeps >=2
    // blocks/CU); otherwise per-step slices as before
    size_t full = sizeof(float) * (2 * a.D + (size_t)a.Smax * a.d +
                                   (size_t)bsmax * a.k) +
                  sizeof(int) * 2 * a.P;
    a.stage_full = (full <= 64 * 1024) ? 1 : 0;
    size_t smem = a.stage_full
        ? full
        : sizeof(float) * (2 * a.D + (size_t)bsmax * a.d +
                           (size_t)bsmax * a.k) +
              sizeof(int) * 2 * a.P;
    TORCH_CHECK(smem <= 160 * 1024, "partitioned logreg LDS budget exceeded: ", smem);
    hipLaunchKernelGGL(tick_logreg_part_kernel, dim3(n), dim3(128), smem,
                       current_stream(), a);
}

void tick_logreg(torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
                 torch::Tensor slot_ages, torch::Tensor nodes, torch::Tensor recv_ptr,
                 torch::Tensor del_slots, torch::Tensor reply_slots,
                 torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
                 int64_t d, int64_t k, double lr, double wd, int64_t epochs,
                 int64_t bs, int64_t mode, bool update_only,
                 torch::Tensor dmodes)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X); CHECK_DEV(Y);
    TORCH_CHECK(k <= KMAX, "n_classes > ", KMAX, " unsupported");
    int n = nodes.size(0);
    if (n == 0) return;
    LogregArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.nodes = nodes.data_ptr<int>(); a.ptr = recv_ptr.data_ptr<int>();
    a.dslots = del_slots.numel() ? del_slots.data_ptr<int>() : nullptr;
    a.rslots = reply_slots.numel() ? reply_slots.data_ptr<int>() : nullptr;
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.dmodes = dmodes.numel() ? dmodes.data_ptr<int>() : nullptr;
    a.d = d; a.k = k; a.Smax = X.size(1); a.D = params.size(1);
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs; a.mode = mode;
    a.update_only = update_only;
    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    size_t smem = sizeof(float) * (2 * a.D + (size_t)bsmax * a.d + (size_t)bsmax * a.k);
    TORCH_CHECK(smem <= 160 * 1024, "logreg LDS budget exceeded: ", smem);
    hipLaunchKernelGGL(tick_logreg_kernel, dim3(n), dim3(128), smem,
                       current_stream(), a);
}

void tick_linear(torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
                 torch::Tensor slot_ages, torch::Tensor nodes, torch::Tensor recv_ptr,
                 torch::Tensor del_slots, torch::Tensor reply_slots,
                 torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
                 int64_t d, double lrlam, int64_t is_pegasos, int64_t mode,
                 bool update_only, torch::Tensor dmodes)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X); CHECK_DEV(Y);
    TORCH_CHECK(d <= LIN_MAX_REGS * WAVE, "d > ", LIN_MAX_REGS * WAVE, " unsupported");
    int n = nodes.size(0);
    if (n == 0) return;
    LinearArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.nodes = nodes.data_ptr<int>(); a.ptr = recv_ptr.data_ptr<int>();
    a.dslots = del_slots.numel() ? del_slots.data_ptr<int>() : nullptr;
    a.rslots = reply_slots.numel() ? reply_slots.data_ptr<int>() : nullptr;
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.dmodes = dmodes.numel() ? dmodes.data_ptr<int>() : nullptr;
    a.d = d; a.Smax = X.size(1);
    a.lrlam = lrlam; a.is_pegasos = is_pegasos; a.mode = mode;
    a.update_only = update_only;
    hipLaunchKernelGGL(tick_linear_kernel, dim3(n), dim3(WAVE), 0,
                       current_stream(), a);
}

void tick_mlp(torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
              torch::Tensor slot_ages, torch::Tensor nodes, torch::Tensor recv_ptr,
              torch::Tensor del_slots, torch::Tensor reply_slots,
              torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
              torch::Tensor layout, int64_t n_layers, double lr, double wd,
              int64_t epochs, int64_t bs, int64_t mode, bool update_only,
              torch::Tensor dmodes)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X); CHECK_DEV(Y);
    CHECK_DEV(layout);
    int n = nodes.size(0);
    if (n == 0) return;
    MlpArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.nodes = nodes.data_ptr<int>(); a.ptr = recv_ptr.data_ptr<int>();
    a.dslots = del_slots.numel() ? del_slots.data_ptr<int>() : nullptr;
    a.rslots = reply_slots.numel() ? reply_slots.data_ptr<int>() : nullptr;
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.layout = layout.data_ptr<int>();
    a.dmodes = dmodes.numel() ? dmodes.data_ptr<int>() : nullptr;
    a.n_layers = n_layers; a.Smax = X.size(1); a.D = params.size(1);
    a.d_in = layout[2].item<int>();
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs; a.mode = mode;
    a.update_only = update_only;
    // LDS: 2*D (model + scratch) + activations + 2 grad buffers
    auto lay = layout.cpu();
    const int* L = lay.data_ptr<int>();
    int act_sum = 0, act_max = L[2];
    for (int l = 0; l < n_layers; ++l) {
        act_sum += L[4 * l + 3];
        act_max = std::max(act_max, L[4 * l + 3]);
        act_max = std::max(act_max, L[4 * l + 2]);
    }
    a.act_max = act_max;
    a.d_in = L[2];
    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    size_t w_slabs = (mode == MODE_UPDATE_MERGE) ? 2 : 1;  // W2 only there
    size_t smem = sizeof(float) *
        (w_slabs * (size_t)a.D + (size_t)bsmax * (a.d_in + act_sum) +
         2 * (size_t)bsmax * act_max);
    TORCH_CHECK(smem <= 160 * 1024,
        "mlp LDS budget exceeded (", smem, " B); shrink batch_size/hidden");
    // block size: 8 waves when any GEMM stage has >= 8 MFMA tiles (halves
    // the sequential tile passes of the widest stage — e.g. dW [100,57] is
    // 28 tiles), else the default 4 waves
    int max_tiles = 0;
    {
        int m16 = (bsmax + 15) >> 4;
        for (int l = 0; l < n_layers; ++l) {
            int fin16 = (L[4 * l + 2] + 15) >> 4;
            int fout16 = (L[4 * l + 3] + 15) >> 4;
            max_tiles = std::max(max_tiles, m16 * fout16);   // forward
            max_tiles = std::max(max_tiles, m16 * fin16);    // dX
            max_tiles = std::max(max_tiles, fout16 * fin16); // dW
        }
    }
    int threads = (max_tiles >= 8) ? 512 : 256;
    hipLaunchKernelGGL(tick_mlp_kernel, dim3(n), dim3(threads), smem,
                       current_stream(), a);
}

// ---------------------------------------------------------------------------
// round executors: the C++ loop over one round's ticks. All event arrays
// arrive on-device ONCE per round; tick boundary pointers stay host-side.
// Per tick this issues up to 4 stream-ordered launches (snapshot, deliver,
// pull-snapshot, reply-deliver) with zero per-tick Python involvement —
// the host cost of a 100-tick round drops from ~10 ms of Python to
// ~200 launch enqueues of a few µs each.
// ---------------------------------------------------------------------------

struct RoundArrays {
    // device pointers
    const int *snap_nodes, *snap_slots;
    const int *recv_nodes, *recv_nptr;
    const int *del_slots, *reply_slots;
    const int *pull_nodes, *pull_slots;
    const int *rep_nodes, *rep_nptr, *rep_slots;
    // host tick pointers [delta+1]
    const int *snap_tptr, *recv_tptr, *pull_tptr, *rep_tptr;
    int delta;
};

static RoundArrays unpack_round(
    torch::Tensor snap_nodes, torch::Tensor snap_slots, torch::Tensor snap_tptr,
    torch::Tensor recv_nodes, torch::Tensor recv_nptr, torch::Tensor recv_tptr,
    torch::Tensor del_slots, torch::Tensor reply_slots,
    torch::Tensor pull_nodes, torch::Tensor pull_slots, torch::Tensor pull_tptr,
    torch::Tensor rep_nodes, torch::Tensor rep_nptr, torch::Tensor rep_tptr,
    torch::Tensor rep_slots)
{
    RoundArrays r;
    r.snap_nodes = snap_nodes.numel() ? snap_nodes.data_ptr<int>() : nullptr;
    r.snap_slots = snap_slots.numel() ? snap_slots.data_ptr<int>() : nullptr;
    r.recv_nodes = recv_nodes.numel() ? recv_nodes.data_ptr<int>() : nullptr;
    r.recv_nptr = recv_nptr.data_ptr<int>();
    r.del_slots = del_slots.numel() ? del_slots.data_ptr<int>() : nullptr;
    r.reply_slots = reply_slots.numel() ? reply_slots.data_ptr<int>() : nullptr;
    r.pull_nodes = pull_nodes.numel() ? pull_nodes.data_ptr<int>() : nullptr;
    r.pull_slots = pull_slots.numel() ? pull_slots.data_ptr<int>() : nullptr;
    r.rep_nodes = rep_nodes.numel() ? rep_nodes.data_ptr<int>() : nullptr;
    r.rep_nptr = rep_nptr.data_ptr<int>();
    r.rep_slots = rep_slots.numel() ? rep_slots.data_ptr<int>() : nullptr;
    r.snap_tptr = snap_tptr.data_ptr<int>();   // host (CPU) tensors
    r.recv_tptr = recv_tptr.data_ptr<int>();
    r.pull_tptr = pull_tptr.data_ptr<int>();
    r.rep_tptr = rep_tptr.data_ptr<int>();
    r.delta = (int)snap_tptr.numel() - 1;
    return r;
}

static int sb_threshold()
{
    // measured on MI355X: the single-workgroup round is ~5% SLOWER than
    // back-to-back stream launches even at 1-3-block batches (the stream
    // front-end pipelines launches well on gfx950), so the path is
    // opt-in via GOSSIPY_SB_MAX (profiles/r01_coop_ab.txt)
    const char* v = getenv("GOSSIPY_SB_MAX");
    return v ? atoi(v) : 0;
}

// largest per-tick batch across all four event kinds (host-side tptrs)
static int max_group(const RoundArrays& r)
{
    int mx = 0;
    for (int t = 0; t < r.delta; ++t) {
        mx = std::max(mx, r.snap_tptr[t + 1] - r.snap_tptr[t]);
        mx = std::max(mx, r.recv_tptr[t + 1] - r.recv_tptr[t]);
        mx = std::max(mx, r.pull_tptr[t + 1] - r.pull_tptr[t]);
        mx = std::max(mx, r.rep_tptr[t + 1] - r.rep_tptr[t]);
    }
    return mx;
}

static void launch_snap(const float* params, const int* ages, float* slots,
                        int* slot_ages, const int* nodes, const int* slot_ids,
                        int n, int D, hipStream_t s, int A = 1)
{
    long total = (long)n * D;
    int block = 256;
    int grid = (int)std::min<long>((total + block - 1) / block, 2048);
    hipLaunchKernelGGL(snapshot_kernel, dim3(grid), dim3(block), 0, s,
                       params, ages, slots, slot_ages, nodes, slot_ids, n, D,
                       D, 0, A);
}

void run_round_logreg(
    torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
    torch::Tensor slot_ages,
    torch::Tensor snap_nodes, torch::Tensor snap_slots, torch::Tensor snap_tptr,
    torch::Tensor recv_nodes, torch::Tensor recv_nptr, torch::Tensor recv_tptr,
    torch::Tensor del_slots, torch::Tensor reply_slots,
    torch::Tensor pull_nodes, torch::Tensor pull_slots, torch::Tensor pull_tptr,
    torch::Tensor rep_nodes, torch::Tensor rep_nptr, torch::Tensor rep_tptr,
    torch::Tensor rep_slots,
    torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
    int64_t d, int64_t k, double lr, double wd, int64_t epochs, int64_t bs,
    int64_t mode,
    torch::Tensor rep_reply_slots)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X);
    RoundArrays r = unpack_round(snap_nodes, snap_slots, snap_tptr, recv_nodes,
                                 recv_nptr, recv_tptr, del_slots, reply_slots,
                                 pull_nodes, pull_slots, pull_tptr, rep_nodes,
                                 rep_nptr, rep_tptr, rep_slots);
    const int* rep_rr =
        (rep_reply_slots.defined() && rep_reply_slots.numel())
            ? rep_reply_slots.data_ptr<int>() : nullptr;
    hipStream_t s = current_stream();
    LogregArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.dmodes = nullptr;
    a.d = d; a.k = k; a.Smax = X.size(1); a.D = params.size(1);
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs; a.mode = mode;
    a.update_only = 0;
    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    size_t smem = sizeof(float) * (2 * a.D + (size_t)bsmax * a.d + (size_t)bsmax * a.k);
    TORCH_CHECK(smem <= 160 * 1024, "logreg LDS budget exceeded");
    if (max_group(r) <= sb_threshold() && rep_rr == nullptr) {
        // tiny batches: the coop round kernel launched PLAIN with one
        // workgroup (round_sync degrades to __syncthreads)
        auto dev = params.device();
        auto st = snap_tptr.to(dev), rt = recv_tptr.to(dev);
        auto pt = pull_tptr.to(dev), qt = rep_tptr.to(dev);
        CoopRoundArgs c;
        c.base = a;
        c.snap_nodes = r.snap_nodes; c.snap_slots = r.snap_slots;
        c.snap_tptr = st.data_ptr<int>();
        c.recv_nodes = r.recv_nodes; c.recv_nptr = r.recv_nptr;
        c.recv_tptr = rt.data_ptr<int>();
        c.del_slots = r.del_slots; c.reply_slots = r.reply_slots;
        c.pull_nodes = r.pull_nodes; c.pull_slots = r.pull_slots;
        c.pull_tptr = pt.data_ptr<int>();
        c.rep_nodes = r.rep_nodes; c.rep_nptr = r.rep_nptr;
        c.rep_tptr = qt.data_ptr<int>();
        c.rep_slots = r.rep_slots;
        c.delta = r.delta;
        hipLaunchKernelGGL(coop_round_logreg_kernel, dim3(1), dim3(128),
                           smem, s, c);
        static thread_local std::vector<torch::Tensor> keep;
        keep.insert(keep.end(), {st, rt, pt, qt});
        if (keep.size() > 64) keep.erase(keep.begin(), keep.begin() + 32);
        return;
    }
    for (int t = 0; t < r.delta; ++t) {
        int s0 = r.snap_tptr[t], s1 = r.snap_tptr[t + 1];
        if (s1 > s0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.snap_nodes + s0, r.snap_slots + s0, s1 - s0, a.D, s);
        int r0 = r.recv_tptr[t], r1 = r.recv_tptr[t + 1];
        if (r1 > r0) {
            a.nodes = r.recv_nodes + r0;
            a.ptr = r.recv_nptr + r0;
            a.dslots = r.del_slots;
            a.rslots = r.reply_slots;
            hipLaunchKernelGGL(tick_logreg_kernel, dim3(r1 - r0), dim3(128),
                               smem, s, a);
        }
        int p0 = r.pull_tptr[t], p1 = r.pull_tptr[t + 1];
        if (p1 > p0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.pull_nodes + p0, r.pull_slots + p0, p1 - p0, a.D, s);
        int q0 = r.rep_tptr[t], q1 = r.rep_tptr[t + 1];
        if (q1 > q0) {
            a.nodes = r.rep_nodes + q0;
            a.ptr = r.rep_nptr + q0;
            a.dslots = r.rep_slots;
            a.rslots = rep_rr;
            hipLaunchKernelGGL(tick_logreg_kernel, dim3(q1 - q0), dim3(128),
                               smem, s, a);
        }
    }
}

void run_round_linear(
    torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
    torch::Tensor slot_ages,
    torch::Tensor snap_nodes, torch::Tensor snap_slots, torch::Tensor snap_tptr,
    torch::Tensor recv_nodes, torch::Tensor recv_nptr, torch::Tensor recv_tptr,
    torch::Tensor del_slots, torch::Tensor reply_slots,
    torch::Tensor pull_nodes, torch::Tensor pull_slots, torch::Tensor pull_tptr,
    torch::Tensor rep_nodes, torch::Tensor rep_nptr, torch::Tensor rep_tptr,
    torch::Tensor rep_slots,
    torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
    int64_t d, double lrlam, int64_t is_pegasos, int64_t mode,
    torch::Tensor rep_reply_slots)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X);
    RoundArrays r = unpack_round(snap_nodes, snap_slots, snap_tptr, recv_nodes,
                                 recv_nptr, recv_tptr, del_slots, reply_slots,
                                 pull_nodes, pull_slots, pull_tptr, rep_nodes,
                                 rep_nptr, rep_tptr, rep_slots);
    const int* rep_rr =
        (rep_reply_slots.defined() && rep_reply_slots.numel())
            ? rep_reply_slots.data_ptr<int>() : nullptr;
    hipStream_t s = current_stream();
    LinearArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.dmodes = nullptr;
    a.d = d; a.Smax = X.size(1);
    a.lrlam = lrlam; a.is_pegasos = is_pegasos; a.mode = mode;
    a.update_only = 0;
    if (max_group(r) <= sb_threshold() && rep_rr == nullptr) {
        // tiny batches: one plain single-workgroup launch for the round
        auto dev = params.device();
        auto st = snap_tptr.to(dev), rt = recv_tptr.to(dev);
        auto pt = pull_tptr.to(dev), qt = rep_tptr.to(dev);
        LinRoundArgs c;
        c.base = a;
        c.snap_nodes = r.snap_nodes; c.snap_slots = r.snap_slots;
        c.snap_tptr = st.data_ptr<int>();
        c.recv_nodes = r.recv_nodes; c.recv_nptr = r.recv_nptr;
        c.recv_tptr = rt.data_ptr<int>();
        c.del_slots = r.del_slots; c.reply_slots = r.reply_slots;
        c.pull_nodes = r.pull_nodes; c.pull_slots = r.pull_slots;
        c.pull_tptr = pt.data_ptr<int>();
        c.rep_nodes = r.rep_nodes; c.rep_nptr = r.rep_nptr;
        c.rep_tptr = qt.data_ptr<int>();
        c.rep_slots = r.rep_slots;
        c.delta = r.delta;
        hipLaunchKernelGGL(sb_round_linear_kernel, dim3(1), dim3(WAVE), 0, s, c);
        // keep the uploaded tptr tensors alive past the async launch
        static thread_local std::vector<torch::Tensor> keep;
        keep.insert(keep.end(), {st, rt, pt, qt});
        if (keep.size() > 64) keep.erase(keep.begin(), keep.begin() + 32);
        return;
    }
    for (int t = 0; t < r.delta; ++t) {
        int s0 = r.snap_tptr[t], s1 = r.snap_tptr[t + 1];
        if (s1 > s0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.snap_nodes + s0, r.snap_slots + s0, s1 - s0, a.d, s);
        int r0 = r.recv_tptr[t], r1 = r.recv_tptr[t + 1];
        if (r1 > r0) {
            a.nodes = r.recv_nodes + r0;
            a.ptr = r.recv_nptr + r0;
            a.dslots = r.del_slots;
            a.rslots = r.reply_slots;
            hipLaunchKernelGGL(tick_linear_kernel, dim3(r1 - r0), dim3(WAVE),
                               0, s, a);
        }
        int p0 = r.pull_tptr[t], p1 = r.pull_tptr[t + 1];
        if (p1 > p0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.pull_nodes + p0, r.pull_slots + p0, p1 - p0, a.d, s);
        int q0 = r.rep_tptr[t], q1 = r.rep_tptr[t + 1];
        if (q1 > q0) {
            a.nodes = r.rep_nodes + q0;
            a.ptr = r.rep_nptr + q0;
            a.dslots = r.rep_slots;
            a.rslots = rep_rr;
            hipLaunchKernelGGL(tick_linear_kernel, dim3(q1 - q0), dim3(WAVE),
                               0, s, a);
        }
    }
}

void tick_logreg_samp(
    torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
    torch::Tensor slot_ages, torch::Tensor nodes, torch::Tensor recv_ptr,
    torch::Tensor del_slots, torch::Tensor reply_slots, torch::Tensor del_seeds,
    torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
    int64_t samp_c, int64_t d, int64_t k, double lr, double wd,
    int64_t epochs, int64_t bs, int64_t mode, bool update_only)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X); CHECK_DEV(Y);
    TORCH_CHECK(k <= KMAX, "n_classes > ", KMAX, " unsupported");
    TORCH_CHECK(mode != MODE_PASS, "Mode PASS not allowed for sampled models.");
    int n = nodes.size(0);
    if (n == 0) return;
    LogregSampArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.nodes = nodes.data_ptr<int>(); a.ptr = recv_ptr.data_ptr<int>();
    a.dslots = del_slots.numel() ? del_slots.data_ptr<int>() : nullptr;
    a.rslots = reply_slots.numel() ? reply_slots.data_ptr<int>() : nullptr;
    a.dseeds = del_seeds.numel() ? del_seeds.data_ptr<int>() : nullptr;
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.samp_c = samp_c; a.d = d; a.k = k; a.Smax = X.size(1);
    a.D = params.size(1);
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs; a.mode = mode;
    a.update_only = update_only;
    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    size_t smem = sizeof(float) * (3 * a.D + (size_t)bsmax * a.d +
                                   (size_t)bsmax * a.k);
    TORCH_CHECK(smem <= 160 * 1024, "sampled logreg LDS budget exceeded: ", smem);
    hipLaunchKernelGGL(tick_logreg_samp_kernel, dim3(n), dim3(128), smem,
                       current_stream(), a);
}

void tick_mf(torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
             torch::Tensor slot_ages, torch::Tensor nodes, torch::Tensor recv_ptr,
             torch::Tensor del_slots, torch::Tensor reply_slots,
             torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
             int64_t k, int64_t n_items, double reg, double lr,
             bool update_only)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X); CHECK_DEV(Y);
    TORCH_CHECK(k <= WAVE, "MF latent dim > ", WAVE, " unsupported");
    int n = nodes.size(0);
    if (n == 0) return;
    MFArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.nodes = nodes.data_ptr<int>(); a.ptr = recv_ptr.data_ptr<int>();
    a.dslots = del_slots.numel() ? del_slots.data_ptr<int>() : nullptr;
    a.rslots = reply_slots.numel() ? reply_slots.data_ptr<int>() : nullptr;
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.k = k; a.n_items = n_items; a.Smax = X.size(1); a.D = params.size(1);
    a.item_off = k + 1; a.Wslot = n_items * (k + 1);
    a.reg = reg; a.lr = lr; a.update_only = update_only;
    hipLaunchKernelGGL(tick_mf_kernel, dim3(n), dim3(256), 0,
                       current_stream(), a);
}

void tick_kmeans(torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
                 torch::Tensor slot_ages, torch::Tensor nodes,
                 torch::Tensor recv_ptr, torch::Tensor del_slots,
                 torch::Tensor reply_slots, torch::Tensor X,
                 torch::Tensor counts, int64_t k, int64_t dim, double alpha,
                 int64_t mode, bool update_only)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X);
    int n = nodes.size(0);
    if (n == 0) return;
    KMeansArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.nodes = nodes.data_ptr<int>(); a.ptr = recv_ptr.data_ptr<int>();
    a.dslots = del_slots.numel() ? del_slots.data_ptr<int>() : nullptr;
    a.rslots = reply_slots.numel() ? reply_slots.data_ptr<int>() : nullptr;
    a.X = X.data_ptr<float>(); a.counts = counts.data_ptr<int>();
    a.k = k; a.dim = dim; a.Smax = X.size(1); a.D = params.size(1);
    a.alpha = alpha; a.mode = mode; a.update_only = update_only;
    size_t smem = sizeof(float) * 2 * a.D + sizeof(int) * a.Smax;
    TORCH_CHECK(smem <= 160 * 1024, "kmeans LDS budget exceeded: ", smem);
    hipLaunchKernelGGL(tick_kmeans_kernel, dim3(n), dim3(128), smem,
                       current_stream(), a);
}

void run_round_logreg_part(
    torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
    torch::Tensor slot_ages,
    torch::Tensor snap_nodes, torch::Tensor snap_slots, torch::Tensor snap_tptr,
    torch::Tensor recv_nodes, torch::Tensor recv_nptr, torch::Tensor recv_tptr,
    torch::Tensor del_slots, torch::Tensor reply_slots, torch::Tensor del_pids,
    torch::Tensor pull_nodes, torch::Tensor pull_slots, torch::Tensor pull_tptr,
    torch::Tensor rep_nodes, torch::Tensor rep_nptr, torch::Tensor rep_tptr,
    torch::Tensor rep_slots, torch::Tensor rep_pids,
    torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
    torch::Tensor perm, torch::Tensor pptr, torch::Tensor apart,
    int64_t n_parts, int64_t d, int64_t k, double lr, double wd,
    int64_t epochs, int64_t bs, int64_t mode,
    torch::Tensor rep_reply_slots)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X);
    CHECK_DEV(perm); CHECK_DEV(pptr); CHECK_DEV(apart);
    TORCH_CHECK(mode != MODE_PASS, "Mode PASS not allowed for partitioned models.");
    RoundArrays r = unpack_round(snap_nodes, snap_slots, snap_tptr, recv_nodes,
                                 recv_nptr, recv_tptr, del_slots, reply_slots,
                                 pull_nodes, pull_slots, pull_tptr, rep_nodes,
                                 rep_nptr, rep_tptr, rep_slots);
    const int* d_pids = del_pids.numel() ? del_pids.data_ptr<int>() : nullptr;
    const int* r_pids = rep_pids.numel() ? rep_pids.data_ptr<int>() : nullptr;
    const int* rep_rr =
        (rep_reply_slots.defined() && rep_reply_slots.numel())
            ? rep_reply_slots.data_ptr<int>() : nullptr;
    hipStream_t s = current_stream();
    LogregPartArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.perm = perm.data_ptr<int>(); a.pptr = pptr.data_ptr<int>();
    a.apart = apart.data_ptr<int>();
    a.P = n_parts; a.d = d; a.k = k; a.Smax = X.size(1); a.D = params.size(1);
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs; a.mode = mode;
    a.update_only = 0;
    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    size_t full = sizeof(float) * (2 * a.D + (size_t)a.Smax * a.d +
                                   (size_t)bsmax * a.k) +
                  sizeof(int) * 2 * a.P;
    a.stage_full = (full <= 64 * 1024) ? 1 : 0;
    size_t smem = a.stage_full
        ? full
        : sizeof(float) * (2 * a.D + (size_t)bsmax * a.d +
                           (size_t)bsmax * a.k) +
              sizeof(int) * 2 * a.P;
    TORCH_CHECK(smem <= 160 * 1024, "partitioned logreg LDS budget exceeded");
    if (max_group(r) <= sb_threshold() && rep_rr == nullptr) {
        auto dev = params.device();
        auto st = snap_tptr.to(dev), rt = recv_tptr.to(dev);
        auto pt = pull_tptr.to(dev), qt = rep_tptr.to(dev);
        PartRoundArgs c;
        c.base = a;
        c.snap_nodes = r.snap_nodes; c.snap_slots = r.snap_slots;
        c.snap_tptr = st.data_ptr<int>();
        c.recv_nodes = r.recv_nodes; c.recv_nptr = r.recv_nptr;
        c.recv_tptr = rt.data_ptr<int>();
        c.del_slots = r.del_slots; c.reply_slots = r.reply_slots;
        c.del_pids = d_pids;
        c.pull_nodes = r.pull_nodes; c.pull_slots = r.pull_slots;
        c.pull_tptr = pt.data_ptr<int>();
        c.rep_nodes = r.rep_nodes; c.rep_nptr = r.rep_nptr;
        c.rep_tptr = qt.data_ptr<int>();
        c.rep_slots = r.rep_slots;
        c.rep_pids = r_pids;
        c.delta = r.delta;
        hipLaunchKernelGGL(sb_round_logreg_part_kernel, dim3(1), dim3(128),
                           smem, s, c);
        static thread_local std::vector<torch::Tensor> keep;
        keep.insert(keep.end(), {st, rt, pt, qt});
        if (keep.size() > 64) keep.erase(keep.begin(), keep.begin() + 32);
        return;
    }
    for (int t = 0; t < r.delta; ++t) {
        int s0 = r.snap_tptr[t], s1 = r.snap_tptr[t + 1];
        if (s1 > s0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.snap_nodes + s0, r.snap_slots + s0, s1 - s0, a.D, s,
                        a.P);
        int r0 = r.recv_tptr[t], r1 = r.recv_tptr[t + 1];
        if (r1 > r0) {
            a.nodes = r.recv_nodes + r0;
            a.ptr = r.recv_nptr + r0;
            a.dslots = r.del_slots;
            a.rslots = r.reply_slots;
            a.dpids = d_pids;
            hipLaunchKernelGGL(tick_logreg_part_kernel, dim3(r1 - r0), dim3(128),
                               smem, s, a);
        }
        int p0 = r.pull_tptr[t], p1 = r.pull_tptr[t + 1];
        if (p1 > p0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.pull_nodes + p0, r.pull_slots + p0, p1 - p0, a.D, s,
                        a.P);
        int q0 = r.rep_tptr[t], q1 = r.rep_tptr[t + 1];
        if (q1 > q0) {
            a.nodes = r.rep_nodes + q0;
            a.ptr = r.rep_nptr + q0;
            a.dslots = r.rep_slots;
            a.rslots = rep_rr;
            a.dpids = r_pids;
            hipLaunchKernelGGL(tick_logreg_part_kernel, dim3(q1 - q0), dim3(128),
                               smem, s, a);
        }
    }
}

void run_round_mlp(
    torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
    torch::Tensor slot_ages,
    torch::Tensor snap_nodes, torch::Tensor snap_slots, torch::Tensor snap_tptr,
    torch::Tensor recv_nodes, torch::Tensor recv_nptr, torch::Tensor recv_tptr,
    torch::Tensor del_slots, torch::Tensor reply_slots,
    torch::Tensor pull_nodes, torch::Tensor pull_slots, torch::Tensor pull_tptr,
    torch::Tensor rep_nodes, torch::Tensor rep_nptr, torch::Tensor rep_tptr,
    torch::Tensor rep_slots,
    torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
    torch::Tensor layout, int64_t n_layers, double lr, double wd,
    int64_t epochs, int64_t bs, int64_t mode,
    torch::Tensor rep_reply_slots)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X); CHECK_DEV(layout);
    RoundArrays r = unpack_round(snap_nodes, snap_slots, snap_tptr, recv_nodes,
                                 recv_nptr, recv_tptr, del_slots, reply_slots,
                                 pull_nodes, pull_slots, pull_tptr, rep_nodes,
                                 rep_nptr, rep_tptr, rep_slots);
    const int* rep_rr =
        (rep_reply_slots.defined() && rep_reply_slots.numel())
            ? rep_reply_slots.data_ptr<int>() : nullptr;
    hipStream_t s = current_stream();
    MlpArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.layout = layout.data_ptr<int>();
    a.dmodes = nullptr;
    a.n_layers = n_layers; a.Smax = X.size(1); a.D = params.size(1);
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs; a.mode = mode;
    a.update_only = 0;
    auto lay = layout.cpu();
    const int* L = lay.data_ptr<int>();
    int act_sum = 0, act_max = L[2];
    for (int l = 0; l < n_layers; ++l) {
        act_sum += L[4 * l + 3];
        act_max = std::max(act_max, L[4 * l + 3]);
        act_max = std::max(act_max, L[4 * l + 2]);
    }
    a.act_max = act_max;
    a.d_in = L[2];
    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    size_t w_slabs = (mode == MODE_UPDATE_MERGE) ? 2 : 1;  // W2 only there
    size_t smem = sizeof(float) *
        (w_slabs * (size_t)a.D + (size_t)bsmax * (a.d_in + act_sum) +
         2 * (size_t)bsmax * act_max);
    TORCH_CHECK(smem <= 160 * 1024, "mlp LDS budget exceeded");
    // 8 waves when any GEMM stage has >= 8 MFMA tiles (same policy as
    // tick_mlp — the packed executor previously hardcoded 4 waves, which
    // left each SIMD with a single wave and every LDS/MFMA stall exposed)
    int max_tiles = 0;
    {
        int m16 = (bsmax + 15) >> 4;
        for (int l = 0; l < n_layers; ++l) {
            int fin16 = (L[4 * l + 2] + 15) >> 4;
            int fout16 = (L[4 * l + 3] + 15) >> 4;
            max_tiles = std::max(max_tiles, m16 * fout16);
            max_tiles = std::max(max_tiles, m16 * fin16);
            max_tiles = std::max(max_tiles, fout16 * fin16);
        }
    }
    int threads = (max_tiles >= 8) ? 512 : 256;
    for (int t = 0; t < r.delta; ++t) {
        int s0 = r.snap_tptr[t], s1 = r.snap_tptr[t + 1];
        if (s1 > s0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.snap_nodes + s0, r.snap_slots + s0, s1 - s0, a.D, s);
        int r0 = r.recv_tptr[t], r1 = r.recv_tptr[t + 1];
        if (r1 > r0) {
            a.nodes = r.recv_nodes + r0;
            a.ptr = r.recv_nptr + r0;
            a.dslots = r.del_slots;
            a.rslots = r.reply_slots;
            hipLaunchKernelGGL(tick_mlp_kernel, dim3(r1 - r0),
                               dim3(threads), smem, s, a);
        }
        int p0 = r.pull_tptr[t], p1 = r.pull_tptr[t + 1];
        if (p1 > p0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.pull_nodes + p0, r.pull_slots + p0, p1 - p0, a.D, s);
        int q0 = r.rep_tptr[t], q1 = r.rep_tptr[t + 1];
        if (q1 > q0) {
            a.nodes = r.rep_nodes + q0;
            a.ptr = r.rep_nptr + q0;
            a.dslots = r.rep_slots;
            a.rslots = rep_rr;
            hipLaunchKernelGGL(tick_mlp_kernel, dim3(q1 - q0),
                               dim3(threads), smem, s, a);
        }
    }
}

void tick_pens(torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
               torch::Tensor slot_ages, torch::Tensor nodes, torch::Tensor ptr,
               torch::Tensor cslots, torch::Tensor cowners,
               torch::Tensor counts, torch::Tensor X, torch::Tensor Y,
               torch::Tensor dcounts, int64_t d, int64_t k, int64_t m_top,
               double lr, double wd, int64_t epochs, int64_t bs)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X); CHECK_DEV(counts);
    TORCH_CHECK(k <= KMAX, "n_classes > ", KMAX, " unsupported");
    int n = nodes.size(0);
    if (n == 0) return;
    PensArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.nodes = nodes.data_ptr<int>(); a.ptr = ptr.data_ptr<int>();
    a.cslots = cslots.data_ptr<int>(); a.cowners = cowners.data_ptr<int>();
    a.counts = counts.data_ptr<int>();
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.dcounts = dcounts.data_ptr<int>();
    a.d = d; a.k = k; a.Smax = X.size(1); a.D = params.size(1);
    a.m_top = m_top; a.n_total = counts.size(1);
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs;
    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    size_t smem = sizeof(float) * (a.D + (size_t)bsmax * a.d +
                                   (size_t)bsmax * a.k);
    TORCH_CHECK(smem <= 160 * 1024, "pens LDS budget exceeded: ", smem);
    hipLaunchKernelGGL(tick_pens_kernel, dim3(n), dim3(128), smem,
                       current_stream(), a);
}

torch::Tensor eval_metrics(torch::Tensor params, torch::Tensor nodes,
                           torch::Tensor X, torch::Tensor Y, int64_t d,
                           int64_t k, bool is_margin,
                           torch::Tensor eval_counts)
{
    CHECK_DEV(params); CHECK_DEV(nodes); CHECK_DEV(X); CHECK_DEV(Y);
    TORCH_CHECK(k <= EVAL_KMAX, "n_classes > ", EVAL_KMAX, " unsupported");
    int R = nodes.size(0);
    bool per_node = eval_counts.defined() && eval_counts.numel();
    int n_eval = per_node ? (int)X.size(1) : (int)X.size(0);
    auto out = torch::empty({R, 5}, params.options());
    if (R == 0) return out;
    EvalArgs a;
    a.params = params.data_ptr<float>();
    a.nodes = nodes.data_ptr<int>();
    a.X = X.data_ptr<float>();
    a.Y = Y.data_ptr<float>();
    a.out = out.data_ptr<float>();
    a.d = d; a.k = k; a.n_eval = n_eval; a.D = params.size(1);
    a.is_margin = is_margin;
    a.scores = nullptr;
    a.eval_counts = per_node ? eval_counts.data_ptr<int>() : nullptr;
    // LDS: model + scores + binarized labels + confusion + one X tile.
    // The tile size adapts so everything fits the 160 KB budget.
    size_t fixed = sizeof(float) * (a.D + n_eval)
        + ((n_eval + 15) & ~15) * sizeof(char)
        + sizeof(int) * EVAL_KMAX * EVAL_KMAX;
    long room = (long)(160 * 1024 - fixed) / (long)(sizeof(float) * d);
    TORCH_CHECK(room >= 16, "eval LDS budget exceeded (n_eval=", n_eval,
                ", d=", d, ")");
    a.tile = (int)std::min<long>({room, (long)n_eval, 512});
    size_t smem = fixed + sizeof(float) * (size_t)a.tile * d;
    hipLaunchKernelGGL(eval_metrics_kernel, dim3(R), dim3(256), smem,
                       current_stream(), a);
    return out;
}

torch::Tensor eval_metrics_scores(torch::Tensor scores, torch::Tensor Y,
                                  int64_t k, bool is_margin)
{
    // K13 on PRECOMPUTED raw scores [R, n_eval, k] ([R, n_eval] margin):
    // families whose forward runs elsewhere (MLP / torchmod) get the same
    // one-launch confusion + macro-PRF + pairwise-AUC epilogue instead of
    // ~40 small torch kernels and a host sync.
    CHECK_DEV(scores); CHECK_DEV(Y);
    TORCH_CHECK(k <= EVAL_KMAX, "n_classes > ", EVAL_KMAX, " unsupported");
    int R = scores.size(0);
    int n_eval = scores.size(1);
    auto out = torch::empty({R, 5}, scores.options());
    if (R == 0) return out;
    EvalArgs a;
    a.params = nullptr; a.nodes = nullptr; a.X = nullptr;
    a.Y = Y.data_ptr<float>();
    a.out = out.data_ptr<float>();
    a.d = 0; a.k = k; a.n_eval = n_eval; a.D = 0;
    a.is_margin = is_margin;
    a.tile = 1;
    a.eval_counts = nullptr;
    a.scores = scores.data_ptr<float>();
    size_t smem = sizeof(float) * n_eval
        + ((n_eval + 15) & ~15) * sizeof(char)
        + sizeof(int) * EVAL_KMAX * EVAL_KMAX;
    TORCH_CHECK(smem <= 160 * 1024, "eval LDS budget exceeded: ", smem);
    hipLaunchKernelGGL(eval_metrics_kernel, dim3(R), dim3(256), smem,
                       current_stream(), a);
    return out;
}

void run_round_logreg_samp(
    torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
    torch::Tensor slot_ages,
    torch::Tensor snap_nodes, torch::Tensor snap_slots, torch::Tensor snap_tptr,
    torch::Tensor recv_nodes, torch::Tensor recv_nptr, torch::Tensor recv_tptr,
    torch::Tensor del_slots, torch::Tensor reply_slots, torch::Tensor del_pids,
    torch::Tensor pull_nodes, torch::Tensor pull_slots, torch::Tensor pull_tptr,
    torch::Tensor rep_nodes, torch::Tensor rep_nptr, torch::Tensor rep_tptr,
    torch::Tensor rep_slots, torch::Tensor rep_pids,
    torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
    int64_t samp_c, int64_t d, int64_t k, double lr, double wd,
    int64_t epochs, int64_t bs, int64_t mode,
    torch::Tensor rep_reply_slots)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X);
    TORCH_CHECK(mode != MODE_PASS, "Mode PASS not allowed for sampled models.");
    RoundArrays r = unpack_round(snap_nodes, snap_slots, snap_tptr, recv_nodes,
                                 recv_nptr, recv_tptr, del_slots, reply_slots,
                                 pull_nodes, pull_slots, pull_tptr, rep_nodes,
                                 rep_nptr, rep_tptr, rep_slots);
    const int* d_pids = del_pids.numel() ? del_pids.data_ptr<int>() : nullptr;
    const int* r_pids = rep_pids.numel() ? rep_pids.data_ptr<int>() : nullptr;
    const int* rep_rr =
        (rep_reply_slots.defined() && rep_reply_slots.numel())
            ? rep_reply_slots.data_ptr<int>() : nullptr;
    hipStream_t s = current_stream();
    LogregSampArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.samp_c = samp_c; a.d = d; a.k = k; a.Smax = X.size(1);
    a.D = params.size(1);
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs; a.mode = mode;
    a.update_only = 0;
    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    size_t smem = sizeof(float) * (3 * a.D + (size_t)bsmax * a.d +
                                   (size_t)bsmax * a.k);
    TORCH_CHECK(smem <= 160 * 1024, "sampled logreg LDS budget exceeded");
    for (int t = 0; t < r.delta; ++t) {
        int s0 = r.snap_tptr[t], s1 = r.snap_tptr[t + 1];
        if (s1 > s0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.snap_nodes + s0, r.snap_slots + s0, s1 - s0, a.D, s);
        int r0 = r.recv_tptr[t], r1 = r.recv_tptr[t + 1];
        if (r1 > r0) {
            a.nodes = r.recv_nodes + r0;
            a.ptr = r.recv_nptr + r0;
            a.dslots = r.del_slots;
            a.rslots = r.reply_slots;
            a.dseeds = d_pids;
            hipLaunchKernelGGL(tick_logreg_samp_kernel, dim3(r1 - r0),
                               dim3(128), smem, s, a);
        }
        int p0 = r.pull_tptr[t], p1 = r.pull_tptr[t + 1];
        if (p1 > p0)
            launch_snap(a.params, a.ages, a.slots, a.slot_ages,
                        r.pull_nodes + p0, r.pull_slots + p0, p1 - p0, a.D, s);
        int q0 = r.rep_tptr[t], q1 = r.rep_tptr[t + 1];
        if (q1 > q0) {
            a.nodes = r.rep_nodes + q0;
            a.ptr = r.rep_nptr + q0;
            a.dslots = r.rep_slots;
            a.rslots = rep_rr;
            a.dseeds = r_pids;
            hipLaunchKernelGGL(tick_logreg_samp_kernel, dim3(q1 - q0),
                               dim3(128), smem, s, a);
        }
    }
}

void run_round_mf(
    torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
    torch::Tensor slot_ages,
    torch::Tensor snap_nodes, torch::Tensor snap_slots, torch::Tensor snap_tptr,
    torch::Tensor recv_nodes, torch::Tensor recv_nptr, torch::Tensor recv_tptr,
    torch::Tensor del_slots, torch::Tensor reply_slots,
    torch::Tensor pull_nodes, torch::Tensor pull_slots, torch::Tensor pull_tptr,
    torch::Tensor rep_nodes, torch::Tensor rep_nptr, torch::Tensor rep_tptr,
    torch::Tensor rep_slots,
    torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
    int64_t k, int64_t n_items, double reg, double lr,
    torch::Tensor rep_reply_slots)
{
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X);
    RoundArrays r = unpack_round(snap_nodes, snap_slots, snap_tptr, recv_nodes,
                                 recv_nptr, recv_tptr, del_slots, reply_slots,
                                 pull_nodes, pull_slots, pull_tptr, rep_nodes,
                                 rep_nptr, rep_tptr, rep_slots);
    const int* rep_rr =
        (rep_reply_slots.defined() && rep_reply_slots.numel())
            ? rep_reply_slots.data_ptr<int>() : nullptr;
    hipStream_t s = current_stream();
    MFArgs a;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.k = k; a.n_items = n_items; a.Smax = X.size(1); a.D = params.size(1);
    a.item_off = k + 1; a.Wslot = n_items * (k + 1);
    a.reg = reg; a.lr = lr; a.update_only = 0;
    // MF snapshots carry only the item block (src_off = item_off)
    for (int t = 0; t < r.delta; ++t) {
        int s0 = r.snap_tptr[t], s1 = r.snap_tptr[t + 1];
        if (s1 > s0) {
            long total = (long)(s1 - s0) * a.Wslot;
            int grid = (int)std::min<long>((total + 255) / 256, 2048);
            hipLaunchKernelGGL(snapshot_kernel, dim3(grid), dim3(256), 0, s,
                               a.params, a.ages, a.slots,
                               slot_ages.data_ptr<int>(), r.snap_nodes + s0,
                               r.snap_slots + s0, s1 - s0, a.Wslot, a.D,
                               a.item_off, 1);
        }
        int r0 = r.recv_tptr[t], r1 = r.recv_tptr[t + 1];
        if (r1 > r0) {
            a.nodes = r.recv_nodes + r0;
            a.ptr = r.recv_nptr + r0;
            a.dslots = r.del_slots;
            a.rslots = r.reply_slots;
            hipLaunchKernelGGL(tick_mf_kernel, dim3(r1 - r0), dim3(256), 0,
                               s, a);
        }
        int p0 = r.pull_tptr[t], p1 = r.pull_tptr[t + 1];
        if (p1 > p0) {
            long total = (long)(p1 - p0) * a.Wslot;
            int grid = (int)std::min<long>((total + 255) / 256, 2048);
            hipLaunchKernelGGL(snapshot_kernel, dim3(grid), dim3(256), 0, s,
                               a.params, a.ages, a.slots,
                               slot_ages.data_ptr<int>(), r.pull_nodes + p0,
                               r.pull_slots + p0, p1 - p0, a.Wslot, a.D,
                               a.item_off, 1);
        }
        int q0 = r.rep_tptr[t], q1 = r.rep_tptr[t + 1];
        if (q1 > q0) {
            a.nodes = r.rep_nodes + q0;
            a.ptr = r.rep_nptr + q0;
            a.dslots = r.rep_slots;
            a.rslots = rep_rr;
            hipLaunchKernelGGL(tick_mf_kernel, dim3(q1 - q0), dim3(256), 0,
                               s, a);
        }
    }
}



void run_round_coop_logreg(
    torch::Tensor params, torch::Tensor ages, torch::Tensor slots,
    torch::Tensor slot_ages,
    torch::Tensor snap_nodes, torch::Tensor snap_slots, torch::Tensor snap_tptr,
    torch::Tensor recv_nodes, torch::Tensor recv_nptr, torch::Tensor recv_tptr,
    torch::Tensor del_slots, torch::Tensor reply_slots,
    torch::Tensor pull_nodes, torch::Tensor pull_slots, torch::Tensor pull_tptr,
    torch::Tensor rep_nodes, torch::Tensor rep_nptr, torch::Tensor rep_tptr,
    torch::Tensor rep_slots,
    torch::Tensor X, torch::Tensor Y, torch::Tensor counts,
    int64_t d, int64_t k, double lr, double wd, int64_t epochs, int64_t bs,
    int64_t mode, int64_t max_batch)
{
    // NOTE: unlike the stream executor, ALL arrays (tick ptrs included)
    // must be device-resident here.
    CHECK_DEV(params); CHECK_DEV(slots); CHECK_DEV(X);
    CHECK_DEV(snap_tptr); CHECK_DEV(recv_tptr); CHECK_DEV(pull_tptr);
    CHECK_DEV(rep_tptr);
    CoopRoundArgs c;
    LogregArgs& a = c.base;
    a.params = params.data_ptr<float>(); a.ages = ages.data_ptr<int>();
    a.slots = slots.data_ptr<float>(); a.slot_ages = slot_ages.data_ptr<int>();
    a.X = X.data_ptr<float>(); a.Y = Y.data_ptr<float>();
    a.counts = counts.data_ptr<int>();
    a.dmodes = nullptr;
    a.d = d; a.k = k; a.Smax = X.size(1); a.D = params.size(1);
    a.lr = lr; a.wd = wd; a.epochs = epochs; a.bs = bs; a.mode = mode;
    a.update_only = 0;
    auto dp = [](torch::Tensor& t) {
        return t.numel() ? t.data_ptr<int>() : nullptr;
    };
    c.snap_nodes = dp(snap_nodes); c.snap_slots = dp(snap_slots);
    c.snap_tptr = snap_tptr.data_ptr<int>();
    c.recv_nodes = dp(recv_nodes); c.recv_nptr = recv_nptr.data_ptr<int>();
    c.recv_tptr = recv_tptr.data_ptr<int>();
    c.del_slots = dp(del_slots); c.reply_slots = dp(reply_slots);
    c.pull_nodes = dp(pull_nodes); c.pull_slots = dp(pull_slots);
    c.pull_tptr = pull_tptr.data_ptr<int>();
    c.rep_nodes = dp(rep_nodes); c.rep_nptr = rep_nptr.data_ptr<int>();
    c.rep_tptr = rep_tptr.data_ptr<int>();
    c.rep_slots = dp(rep_slots);
    c.delta = (int)snap_tptr.numel() - 1;

    int bsmax = (bs == 0) ? a.Smax : std::min<int>(bs, a.Smax);
    size_t smem = sizeof(float) * (2 * a.D + (size_t)bsmax * a.d +
                                   (size_t)bsmax * a.k);
    TORCH_CHECK(smem <= 160 * 1024, "logreg LDS budget exceeded");
    int grid = std::max<int>(1, (int)max_batch);
    // stay within guaranteed cooperative residency
    int max_blocks = 0;
    hipError_t oerr = hipOccupancyMaxActiveBlocksPerMultiprocessor(
        &max_blocks, (const void*)coop_round_logreg_kernel, 128, smem);
    TORCH_CHECK(oerr == hipSuccess, "occupancy query failed");
    hipDeviceProp_t prop;
    (void)hipGetDeviceProperties(&prop, 0);
    int limit = max_blocks * prop.multiProcessorCount;
    TORCH_CHECK(limit > 0, "cooperative launch not supported");
    if (grid > limit) grid = limit;
    void* kargs[] = {&c};
    hipError_t err = hipLaunchCooperativeKernel(
        (const void*)coop_round_logreg_kernel, dim3(grid), dim3(128), kargs,
        smem, current_stream());
    TORCH_CHECK(err == hipSuccess, "cooperative launch failed: ",
                hipGetErrorString(err));
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m)
{
    m.def("snapshot", &snapshot, "batched model snapshot (arena row copy)");
    m.def("tick_logreg", &tick_logreg, "fused merge + logreg SGD tick");
    m.def("tick_linear", &tick_linear, "fused merge + pegasos/adaline tick");
    m.def("tick_mlp", &tick_mlp, "fused merge + MLP SGD tick");
    m.def("run_round_logreg", &run_round_logreg,
          "whole-round executor, logreg family",
          py::call_guard<py::gil_scoped_release>());
    m.def("run_round_linear", &run_round_linear,
          "whole-round executor, pegasos/adaline family",
          py::call_guard<py::gil_scoped_release>());
    m.def("tick_logreg_part", &tick_logreg_part,
          "fused partition-merge + age-rescaled logreg SGD tick (K7/K8)");
    m.def("run_round_logreg_part", &run_round_logreg_part,
          "whole-round executor, partitioned logreg family",
          py::call_guard<py::gil_scoped_release>());
    m.def("wmerge", &wmerge, "all2all weighted k-way merge (K5 weighted)");
    m.def("tick_logreg_samp", &tick_logreg_samp,
          "fused sampled-merge + logreg SGD tick (K6)");
    m.def("tick_mf", &tick_mf,
          "fused item-block merge + per-rating MF SGD tick (K9/K10)");
    m.def("run_round_mlp", &run_round_mlp,
          "whole-round executor, MLP family",
          py::call_guard<py::gil_scoped_release>());
    m.def("run_round_logreg_samp", &run_round_logreg_samp,
          "whole-round executor, sampled logreg family",
          py::call_guard<py::gil_scoped_release>());
    m.def("run_round_mf", &run_round_mf,
          "whole-round executor, MF recommender family",
          py::call_guard<py::gil_scoped_release>());
    m.def("run_round_coop_logreg", &run_round_coop_logreg,
          "single-launch cooperative whole-round executor (logreg)");
    m.def("tick_pens", &tick_pens,
          "PENS step-1 event: score candidates, merge top-m, count winners");
    m.def("eval_metrics", &eval_metrics,
          "fused per-node eval metrics on a shared eval set (K13)");
    m.def("eval_metrics_scores", &eval_metrics_scores,
          "K13 metrics epilogue on precomputed scores (MLP/torchmod)");
    m.def("tick_kmeans", &tick_kmeans,
          "fused centroid merge + assign/EMA k-means tick (K11)");
}
