// acg_amd gfx950 (CDNA4 / MI355X) kernels for distributed conjugate gradient.
//
// Hand-written HIP replacing the reference's hipSPARSE/hipBLAS calls and its
// CUDA-era kernels (reference: acg/cg-kernels-hip.hip, acg/halo-kernels-hip.hip,
// acg/cghip.c:463-585).  Contents, in file order:
//   1. deterministic block reductions (partials + finalize; no fp64 atomics:
//      a 3000-block atomicAdd onto one cacheline measured 74 us for a 25 MB
//      dot on this chip, vs ~8 us for this scheme -- the reference's
//      unsafeAtomicAdd dots, cg-kernels-hip.hip:1229-1286, are both slower
//      and non-deterministic),
//   2. sparse operators, chosen per matrix:
//      * CSR vector kernel (4-64 lanes/row) for irregular rows and matO,
//      * SELL-C-64 / sigma-SELL (sliced ELLPACK, slice = one 64-lane wave):
//        one row per lane, vals/cols column-major per slice so every wave
//        load is a contiguous 512 B line set; NT loads + unroll-8 variants,
//      * Block-SELL for dense dof x dof block structure (FEM): one int32
//        index per block, pair-major values (16 B/lane dwordx4 loads),
//   3. fused CG kernels with device-resident scalars (alpha/beta computed
//      in-kernel from the 8-slot slab; the only per-iteration D2H is the
//      8-byte convergence norm): classic fused update + combined finalize,
//      pipelined 6-vector update with BOTH next-iteration dots fused in,
//      and the megafused single-kernel iteration (SpMV + update + dots,
//      double-buffered w, split matA/matO passes for halo overlap),
//   4. on-GPU stencil-operator generation (SELL/BSELL built directly in
//      HBM at write speed; enables the 288 GB-per-GPU Poisson sizing),
//   5. the monolithic device-side CG (whole solve in ONE cooperative
//      launch) with a hand-rolled agent-scope grid barrier (sc1
//      write-through payloads, per-XCD generation lines, parity counters,
//      bounded spins -- cooperative grid.sync measured ~150 us/sync),
//   6. halo pack gather (no unpack exists: ghosts are received in place).
//
// The scalar slab layout (fp64 slots) is shared with solvers/hip.py:
#define S_RR 0         // (r,r) current
#define S_PT 1         // (p,t)
#define S_RR_PREV 2    // (r,r) previous
#define S_BNRM2 3      // (b,b)
#define S_GAMMA 4      // pipelined (r,r);  GAMMA,DELTA adjacent => ONE
#define S_DELTA 5      // pipelined (w,r)   2-double allreduce per iteration
#define S_GAMMA_PREV 6
#define S_ALPHA_PREV 7
#define S_NSLOTS 8

// partials scratch: [0, MAXG) first accumulator, [MAXG, 2*MAXG) second
#define MAXG 16384

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>
#include <array>
#include <cstdint>
#include <vector>
#include <stdexcept>
#include <string>

namespace py = pybind11;

#define WAVE 64
#define BLOCK 256

static inline void check_hip(const char* what) {
    hipError_t e = hipGetLastError();
    if (e != hipSuccess)
        throw std::runtime_error(std::string("HIP error in ") + what + ": " + hipGetErrorString(e));
}

static inline long elem_grid(long n, long per_thread = 4) {
    long blocks = (n + (long)BLOCK * per_thread - 1) / ((long)BLOCK * per_thread);
    if (blocks > 4096) blocks = 4096;  // memory-bound: cap + grid-stride (guide §6 G11)
    if (blocks < 1) blocks = 1;
    return blocks;
}

// ---------------------------------------------------------------------------
// block reduction -> one partial per block (deterministic, no atomics)
__device__ __forceinline__ double block_reduce(double v) {
    #pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        v += __shfl_down(v, off, WAVE);
    __shared__ double w[BLOCK / WAVE];
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x >> 6;
    if (lane == 0) w[wid] = v;
    __syncthreads();
    if (wid == 0) {
        v = (lane < BLOCK / WAVE) ? w[lane] : 0.0;
        #pragma unroll
        for (int off = (BLOCK / WAVE) / 2; off > 0; off >>= 1)
            v += __shfl_down(v, off, WAVE);
    }
    return v;  // valid in thread 0
}

// 0/0-safe coefficient division: once the recursion residual underflows
// to EXACT zero (a converged solve driven past convergence, e.g. a
// fixed-iteration benchmark with rtol=0 -- measured: Queen rr reaches
// 3e-136 after 500 iterations and would hit 0 near ~1200), alpha/beta
// become 0/0 = NaN and poison every vector.  Dividing to 0 instead
// freezes the (already converged) iterate, which is exactly the right
// no-op behaviour.
__device__ __forceinline__ double safe_div(double a, double b) {
    return b != 0.0 ? a / b : 0.0;
}

// sum partials[0..nblocks) into scal[slot] (+= if ACC)
__global__ void __launch_bounds__(BLOCK)
k_reduce_partials(const double* __restrict__ partials, int nblocks,
                  double* scal, int slot, int accumulate) {
    double v = 0.0;
    for (int i = threadIdx.x; i < nblocks; i += BLOCK) v += partials[i];
    v = block_reduce(v);
    if (threadIdx.x == 0) scal[slot] = accumulate ? scal[slot] + v : v;
}

// ---------------------------------------------------------------------------
// CSR SpMV, vector kernel: LANES lanes cooperate on one row (irregular rows,
// matO).  FUSE_DOT: also reduce dot(x, y_contrib) into partials[blockIdx].
// ROWLIST: indirect row ids (rowlist[r]) -- the row-binned hybrid launches
// this kernel once per length bin with bin-appropriate LANES, the MI355X
// answer to the load-balance problem the reference's merge-path SpMV
// solves (cg-kernels-hip.hip:348-1175): per-row work stays proportional
// to row length, no fp64 atomics, fully deterministic.
template <typename ColT, int LANES, bool ACCUM, bool FUSE_DOT, bool ROWLIST = false>
__global__ void __launch_bounds__(BLOCK)
spmv_csr_vector(long nrows, long rowbase,
                const long* __restrict__ rowptr,
                const ColT* __restrict__ colidx,
                const double* __restrict__ vals,
                const double* __restrict__ x,
                double* __restrict__ y,
                double* __restrict__ partials,
                const int* __restrict__ rowlist = nullptr) {
    const int lane = threadIdx.x & (LANES - 1);
    const long group = ((long)blockIdx.x * BLOCK + threadIdx.x) / LANES;
    const long ngroups = (long)gridDim.x * BLOCK / LANES;
    double dacc = 0.0;
    for (long i = group; i < nrows; i += ngroups) {
        const long r = ROWLIST ? (long)rowlist[i] : i;
        const long k0 = rowptr[r], k1 = rowptr[r + 1];
        double sum = 0.0;
        for (long k = k0 + lane; k < k1; k += LANES)
            sum += vals[k] * x[colidx[k]];
        #pragma unroll
        for (int off = LANES / 2; off > 0; off >>= 1)
            sum += __shfl_down(sum, off, LANES);
        if (lane == 0) {
            const long row = rowbase + r;
            if (ACCUM) y[row] += sum; else y[row] = sum;
            if (FUSE_DOT) dacc += x[row] * sum;
        }
    }
    if (FUSE_DOT) {
        dacc = block_reduce(dacc);
        if (threadIdx.x == 0) partials[blockIdx.x] = dacc;
    }
}

// ---------------------------------------------------------------------------
// SELL-C-64 SpMV: rows grouped in 64-row slices, vals/cols column-major per
// slice (element j of row (s*64+lane) at sellptr[s] + j*64 + lane).  One wave
// per slice: lane = row; every load is a contiguous 64-lane line; per-row sum
// stays in-register (no cross-lane reduce, no pointer walk).
// Variant knobs (bitmask "variant" on the host API, A/B-tested on MI355X):
//   NT : non-temporal loads for vals/cols (streamed exactly once per SpMV;
//        no-allocate keeps the resident x vector from being evicted from
//        L2 by the 4 GB vals/cols stream)
//   SWZ: XCD-aware slice assignment.  The dispatcher places block b on XCD
//        b%8 (guide §1); mapping contiguous slice ranges to one XCD makes
//        each XCD's x working set ~1/8 of x (~4 MB at Queen scale = its L2).
#define SELL_NT 1
#define SELL_SWZ 2
#define SELL_U8 4

template <typename T>
__device__ __forceinline__ T ld_nt(const T* p) { return __builtin_nontemporal_load(p); }

template <typename ColT, bool ACCUM, bool FUSE_DOT, bool NT, bool SWZ, int UNROLL,
          bool PERM = false>
__global__ void __launch_bounds__(BLOCK)
k_spmv_sell(long nslices, long nrows, long rowbase,
          const long* __restrict__ sellptr,   // [nslices+1], element offsets
          const ColT* __restrict__ cols,      // padded entries: col of pad = row
          const double* __restrict__ vals,    // pad value = 0
          const double* __restrict__ x,
          double* __restrict__ y,
          double* __restrict__ partials,
          const int* __restrict__ perm = nullptr) {  // SELL row -> matrix row (sigma-sorted)
    const int lane = threadIdx.x & (WAVE - 1);
    long blk = blockIdx.x;
    if (SWZ) {
        // bijective XCD remap of the BLOCK index (dispatcher places block b
        // on XCD b%8): XCD k then owns a contiguous slice range, so its x
        // working set is ~1/8 of x and fits the per-XCD 4 MB L2.
        const long nb = gridDim.x, q = nb / 8, rr = nb % 8, xcd = blk % 8;
        blk = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + blk / 8;
    }
    const long wslice = (blk * BLOCK + threadIdx.x) >> 6;
    const long nw = ((long)gridDim.x * BLOCK) >> 6;
    double dacc = 0.0;
    for (long s = wslice; s < nslices; s += nw) {
        const long base = sellptr[s];
        const long len = (sellptr[s + 1] - base) >> 6;  // entries per row
        const double* __restrict__ v = vals + base + lane;
        const ColT* __restrict__ c = cols + base + lane;
        double sum = 0.0;
        long j = 0;
        for (; j + UNROLL <= len; j += UNROLL) {
            double a[UNROLL], xx[UNROLL];
            #pragma unroll
            for (int u = 0; u < UNROLL; ++u) {
                a[u] = NT ? ld_nt(v + (j + u) * WAVE) : v[(j + u) * WAVE];
                const ColT ci = NT ? ld_nt(c + (j + u) * WAVE) : c[(j + u) * WAVE];
                xx[u] = x[ci];
            }
            #pragma unroll
            for (int u = 0; u < UNROLL; ++u) sum += a[u] * xx[u];
        }
        for (; j < len; ++j) {
            const double a = NT ? ld_nt(v + j * WAVE) : v[j * WAVE];
            const ColT ci = NT ? ld_nt(c + j * WAVE) : c[j * WAVE];
            sum += a * x[ci];
        }
        const long row = PERM ? (long)perm[s * WAVE + lane] : s * WAVE + lane;
        if (row < nrows) {
            if (ACCUM) y[rowbase + row] += sum; else y[rowbase + row] = sum;
            if (FUSE_DOT) dacc += x[rowbase + row] * sum;
        }
    }
    if (FUSE_DOT) {
        dacc = block_reduce(dacc);
        if (threadIdx.x == 0) partials[blockIdx.x] = dacc;
    }
}

// ---------------------------------------------------------------------------
// Block-SELL (BSELL): SELL over BLOCK-rows for matrices with a dense
// dof x dof block structure (FEM/structural problems like Queen_4147:
// 3 dof per mesh node).  One lane = one block-row (node); per block only
// ONE int32 block-column index covers dof*dof values, cutting index
// traffic from 4 B/nnz to 4/dof^2 B/nnz (Queen dof=3: 12 -> 8.44 B/nnz
// total, ~30% less SpMV traffic -- the iteration is at the HBM roofline,
// so bytes are the only lever left).
//
// Layout per 64-node slice s (bptr in block units):
//   bcol[bptr[s] + j*64 + lane]              block-col of node's j-th block
//   bvals: PAIR-major so each lane's value pair (k, k+1) is 16 B
//   contiguous and clang emits dwordx4 loads (16 B/lane = the coalescing
//   sweet spot): element (j,k,lane) at
//     bptr[s]*dof^2 + j*dof^2*64 + bval_off(k, lane)
//   with bval_off = (k/2)*128 + lane*2 + (k&1) for paired k, and the odd
//   tail element (k = dof^2-1 when dof^2 is odd) at (dof^2-1)*64 + lane.
__device__ __forceinline__ long stencil_col_node(
    int xi, int yi, int zi, int dx, int dy, int dz, int gx, int gy, int gz,
    const long* __restrict__ pb);  // defined with the stencil generators below

template <int D2>
__device__ __forceinline__ long bval_off(int k, int lane) {
    constexpr int EVEN = D2 & ~1;
    return (k < EVEN) ? (long)(k >> 1) * (2 * WAVE) + lane * 2 + (k & 1)
                      : (long)EVEN * WAVE + lane;
}

template <int DOF, bool FUSE_DOT>
__global__ void __launch_bounds__(BLOCK)
k_spmv_bsell(long nslices, long nnodes,
             const long* __restrict__ bptr,
             const int* __restrict__ bcol,
             const double* __restrict__ bvals,
             const double* __restrict__ x,
             double* __restrict__ y,
             double* __restrict__ partials) {
    const int lane = threadIdx.x & (WAVE - 1);
    const long wslice = ((long)blockIdx.x * BLOCK + threadIdx.x) >> 6;
    const long nw = ((long)gridDim.x * BLOCK) >> 6;
    double dacc = 0.0;
    for (long s = wslice; s < nslices; s += nw) {
        const long b0 = bptr[s];
        const long blen = (bptr[s + 1] - b0) >> 6;  // blocks per node
        const int* __restrict__ c = bcol + b0 + lane;
        const double* __restrict__ v = bvals + b0 * (DOF * DOF);
        double acc[DOF];
        #pragma unroll
        for (int r = 0; r < DOF; ++r) acc[r] = 0.0;
        for (long j = 0; j < blen; ++j) {
            const int cb = ld_nt(c + j * WAVE);
            double xv[DOF];
            #pragma unroll
            for (int cc = 0; cc < DOF; ++cc) xv[cc] = x[(long)cb * DOF + cc];
            const double* __restrict__ vj = v + j * (DOF * DOF) * WAVE;
            #pragma unroll
            for (int k = 0; k < DOF * DOF; ++k) {
                const double a = ld_nt(vj + bval_off<DOF * DOF>(k, lane));
                acc[k / DOF] += a * xv[k % DOF];
            }
        }
        const long node = s * WAVE + lane;
        if (node < nnodes) {
            #pragma unroll
            for (int r = 0; r < DOF; ++r) {
                y[node * DOF + r] = acc[r];
                if (FUSE_DOT) dacc += x[node * DOF + r] * acc[r];
            }
        }
    }
    if (FUSE_DOT) {
        dacc = block_reduce(dacc);
        if (threadIdx.x == 0) partials[blockIdx.x] = dacc;
    }
}

// Classic-CG daypx folded into the BSELL SpMV (serial matA-only path):
// instead of a separate p = beta p + r kernel (3n streams) the gather
// computes beta*p_old[c] + r[c] on the fly and the row side materialises
// p_new = beta*p_old + r into a SEPARATE buffer (ping-pong: folding into
// one buffer would be a write-after-read race against other blocks'
// gathers -- the separate daypx kernel's global barrier is exactly what
// a single buffer needs).  beta = rr/rr_prev from the device scalar
// slab (the previous update's finalize); the host seeds rr_prev = inf
// before iteration 0 so beta = 0 reproduces p0 = r0.  The fused (p,t)
// dot uses the in-register p_new row values.
template <int DOF>
__global__ void __launch_bounds__(BLOCK)
k_spmv_bsell_daypx(long nslices, long nnodes,
                   const long* __restrict__ bptr,
                   const int* __restrict__ bcol,
                   const double* __restrict__ bvals,
                   const double* __restrict__ pold,
                   const double* __restrict__ rvec,
                   double* __restrict__ pnew,
                   double* __restrict__ y,
                   const double* __restrict__ scal,
                   double* __restrict__ partials) {
    const int lane = threadIdx.x & (WAVE - 1);
    const long wslice = ((long)blockIdx.x * BLOCK + threadIdx.x) >> 6;
    const long nw = ((long)gridDim.x * BLOCK) >> 6;
    const double beta = safe_div(scal[S_RR], scal[S_RR_PREV]);
    double dacc = 0.0;
    for (long s = wslice; s < nslices; s += nw) {
        const long b0 = bptr[s];
        const long blen = (bptr[s + 1] - b0) >> 6;
        const int* __restrict__ c = bcol + b0 + lane;
        const double* __restrict__ v = bvals + b0 * (DOF * DOF);
        double acc[DOF];
        #pragma unroll
        for (int r = 0; r < DOF; ++r) acc[r] = 0.0;
        for (long j = 0; j < blen; ++j) {
            const int cb = ld_nt(c + j * WAVE);
            double xv[DOF];
            #pragma unroll
            for (int cc = 0; cc < DOF; ++cc)
                xv[cc] = beta * pold[(long)cb * DOF + cc]
                         + rvec[(long)cb * DOF + cc];
            const double* __restrict__ vj = v + j * (DOF * DOF) * WAVE;
            #pragma unroll
            for (int k = 0; k < DOF * DOF; ++k) {
                const double a = ld_nt(vj + bval_off<DOF * DOF>(k, lane));
                acc[k / DOF] += a * xv[k % DOF];
            }
        }
        const long node = s * WAVE + lane;
        if (node < nnodes) {
            #pragma unroll
            for (int r = 0; r < DOF; ++r) {
                const double pn = beta * pold[node * DOF + r]
                                  + rvec[node * DOF + r];
                pnew[node * DOF + r] = pn;
                y[node * DOF + r] = acc[r];
                dacc += pn * acc[r];
            }
        }
    }
    dacc = block_reduce(dacc);
    if (threadIdx.x == 0) partials[blockIdx.x] = dacc;
}

// device-side BSELL generation for the block-stencil slab (block-level
// analog of k_stencil_rowlen / k_stencil_fill; matA only -- owned x owned)
__global__ void __launch_bounds__(BLOCK)
k_stencil_blocklen(long nnodes, int gx, int gy, int gz, long nown_nodes,
                   const int* __restrict__ zs_of_plane,
                   const long* __restrict__ pb,
                   const double* __restrict__ offs, int ksten,
                   long* __restrict__ blocklen) {
    const long stride = (long)gridDim.x * BLOCK;
    const long plane_nodes = (long)gx * gy;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nnodes; i += stride) {
        const int pl = (int)(i / plane_nodes);
        const long rem = i - (long)pl * plane_nodes;
        const int xi = (int)(rem % gx), yi = (int)(rem / gx);
        const int zi = zs_of_plane[pl];
        long cnt = 1;  // self block
        for (int o = 0; o < ksten; ++o) {
            const long cn = stencil_col_node(xi, yi, zi, (int)offs[o * 4],
                                             (int)offs[o * 4 + 1],
                                             (int)offs[o * 4 + 2], gx, gy, gz, pb);
            if (cn >= 0 && cn < nown_nodes) ++cnt;
        }
        blocklen[i] = cnt;
    }
}

template <int DOF>
__global__ void __launch_bounds__(BLOCK)
k_stencil_bfill(long nnodes, int gx, int gy, int gz, long nown_nodes,
                const int* __restrict__ zs_of_plane,
                const long* __restrict__ pb,
                const double* __restrict__ offs, int ksten,
                const double* __restrict__ blocks,  // M then D
                const long* __restrict__ bptr,
                int* __restrict__ bcol, double* __restrict__ bvals) {
    const long stride = (long)gridDim.x * BLOCK;
    const long plane_nodes = (long)gx * gy;
    const double* M = blocks;
    const double* D = blocks + DOF * DOF;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nnodes; i += stride) {
        const int pl = (int)(i / plane_nodes);
        const long rem = i - (long)pl * plane_nodes;
        const int xi = (int)(rem % gx), yi = (int)(rem / gx);
        const int zi = zs_of_plane[pl];
        const long s = i >> 6;
        const int lane = (int)(i & 63);
        const long b0 = bptr[s];
        const long blen = (bptr[s + 1] - b0) >> 6;
        long j = 0;
        double* __restrict__ bv = bvals + b0 * DOF * DOF;
        // self block (D)
        bcol[b0 + j * WAVE + lane] = (int)i;
        for (int k = 0; k < DOF * DOF; ++k)
            bv[j * DOF * DOF * WAVE + bval_off<DOF * DOF>(k, lane)] = D[k];
        ++j;
        for (int o = 0; o < ksten; ++o) {
            const long cn = stencil_col_node(xi, yi, zi, (int)offs[o * 4],
                                             (int)offs[o * 4 + 1],
                                             (int)offs[o * 4 + 2], gx, gy, gz, pb);
            if (cn < 0 || cn >= nown_nodes) continue;
            const double w = offs[o * 4 + 3];
            bcol[b0 + j * WAVE + lane] = (int)cn;
            for (int k = 0; k < DOF * DOF; ++k)
                bv[j * DOF * DOF * WAVE + bval_off<DOF * DOF>(k, lane)] = w * M[k];
            ++j;
        }
        for (; j < blen; ++j) {  // padding: self col, zero block
            bcol[b0 + j * WAVE + lane] = (int)i;
            for (int k = 0; k < DOF * DOF; ++k)
                bv[j * DOF * DOF * WAVE + bval_off<DOF * DOF>(k, lane)] = 0.0;
        }
    }
}

// ---------------------------------------------------------------------------
// BLAS-1 / fused CG kernels.  Scalar coefficients come from the device slab.

__global__ void __launch_bounds__(BLOCK)
k_zero_scalars(double* scal, int i0, int count) {
    for (int i = threadIdx.x; i < count; i += BLOCK) scal[i0 + i] = 0.0;
}

// before the classic halo/SpMV: zero the (p,t) accumulator (only needed
// on the CSR fallback path; the SELL/BSELL matA finalize overwrites)
__global__ void k_cg_prep_pt(double* scal) { if (threadIdx.x == 0) scal[S_PT] = 0.0; }

// after allreduce(p,t): save rr for the device-side alpha
__global__ void k_cg_prep_rr(double* scal) {
    if (threadIdx.x == 0) scal[S_RR_PREV] = scal[S_RR];
}

// classic-iteration epilogue: rotate rr -> rr_prev, then publish the new
// (r,r) from the update kernel's partials (one 1-block launch instead of
// prep_rr + reduce_partials)
__global__ void __launch_bounds__(BLOCK)
k_cg_finalize(const double* __restrict__ partials, int nblocks, double* scal) {
    double v = 0.0;
    for (int i = threadIdx.x; i < nblocks; i += BLOCK) v += partials[i];
    v = block_reduce(v);
    if (threadIdx.x == 0) {
        scal[S_RR_PREV] = scal[S_RR];
        scal[S_RR] = v;
    }
}

// Jacobi-PCG fused iteration epilogue (beyond reference -- aCG has no
// preconditioning): alpha = rz/(p,t) from the device slab, then ONE pass
// computes r -= alpha t, x += alpha p, z = dinv .* r, and BOTH next
// scalars rz' = (r,z) and rr' = (r,r).  Replaces axpy_ratio x2 + mul +
// prep + dot x2 (6 launches, ~13n traffic) with 1 launch + a finalize.
__global__ void __launch_bounds__(BLOCK)
k_pcg_fused_update(double* __restrict__ r, double* __restrict__ x,
                   const double* __restrict__ p, const double* __restrict__ t,
                   double* __restrict__ z, const double* __restrict__ dinv,
                   long n, const double* __restrict__ scal,
                   double* __restrict__ partials) {
    const double alpha = safe_div(scal[S_RR], scal[S_PT]);  // S_RR = rz
    double arz = 0.0, arr = 0.0;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
        const double rn = r[i] - alpha * ld_nt(t + i);
        r[i] = rn;
        __builtin_nontemporal_store(ld_nt(x + i) + alpha * p[i], x + i);
        const double zi = dinv[i] * rn;
        z[i] = zi;
        arz += rn * zi;
        arr += rn * rn;
    }
    arz = block_reduce(arz);
    __syncthreads();  // block_reduce reuses its LDS scratch
    arr = block_reduce(arr);
    if (threadIdx.x == 0) {
        partials[blockIdx.x] = arz;
        partials[MAXG + blockIdx.x] = arr;
    }
}

// finalize: rotate rz -> rz_prev (S_RR -> S_RR_PREV), publish rz' in
// S_RR and the true rr' in S_GAMMA
__global__ void __launch_bounds__(BLOCK)
k_pcg_finalize(const double* __restrict__ partials, int nblocks,
               double* scal) {
    double vz = 0.0, vr = 0.0;
    for (int i = threadIdx.x; i < nblocks; i += BLOCK) {
        vz += partials[i];
        vr += partials[MAXG + i];
    }
    vz = block_reduce(vz);
    __syncthreads();
    vr = block_reduce(vr);
    if (threadIdx.x == 0) {
        scal[S_RR_PREV] = scal[S_RR];
        scal[S_RR] = vz;
        scal[S_GAMMA] = vr;
    }
}

// dot: partials[b] = block sum of x[i]*y[i]
__global__ void __launch_bounds__(BLOCK)
k_dot(const double* __restrict__ x, const double* __restrict__ y, long n,
      double* __restrict__ partials) {
    double acc = 0.0;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride)
        acc += x[i] * y[i];
    acc = block_reduce(acc);
    if (threadIdx.x == 0) partials[blockIdx.x] = acc;
}

// fused dot2 (pipelined init): gamma_b = sum r*r, delta_b = sum w*r
__global__ void __launch_bounds__(BLOCK)
k_dot2(const double* __restrict__ r, const double* __restrict__ w, long n,
       double* __restrict__ partials) {
    double g = 0.0, d = 0.0;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
        const double ri = r[i];
        g += ri * ri;
        d += w[i] * ri;
    }
    g = block_reduce(g);
    __syncthreads();
    d = block_reduce(d);
    if (threadIdx.x == 0) {
        partials[blockIdx.x] = g;
        partials[MAXG + blockIdx.x] = d;
    }
}

// sum both dot2 partial sets into scal[GAMMA], scal[DELTA]
__global__ void __launch_bounds__(BLOCK)
k_reduce_partials2(const double* __restrict__ partials, int nblocks,
                   double* scal, int accumulate) {
    double g = 0.0, d = 0.0;
    for (int i = threadIdx.x; i < nblocks; i += BLOCK) {
        g += partials[i];
        d += partials[MAXG + i];
    }
    g = block_reduce(g);
    __syncthreads();
    d = block_reduce(d);
    if (threadIdx.x == 0) {
        scal[S_GAMMA] = accumulate ? scal[S_GAMMA] + g : g;
        scal[S_DELTA] = accumulate ? scal[S_DELTA] + d : d;
    }
}

// y += sign * (scal[num]/scal[den]) * x
__global__ void __launch_bounds__(BLOCK)
k_axpy_ratio(double* __restrict__ y, const double* __restrict__ x, long n,
             const double* __restrict__ scal, int num, int den, double sign) {
    const double a = sign * safe_div(scal[num], scal[den]);
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride)
        y[i] += a * x[i];
}

// y = (scal[num]/scal[den]) * y + x      (daypx with device beta)
__global__ void __launch_bounds__(BLOCK)
k_daypx_ratio(double* __restrict__ y, const double* __restrict__ x, long n,
              const double* __restrict__ scal, int num, int den) {
    const double b = safe_div(scal[num], scal[den]);
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride)
        y[i] = b * y[i] + x[i];
}

// classic-CG fused update: alpha = rr/pt (device; S_RR still holds the
// current rr -- k_cg_finalize rotates it afterwards);
//   r -= alpha*t;  x += alpha*p;  partials[b] = block sum of new r.r
// One pass over r,t,x,p instead of three kernels + a dot
// (reference: daxpy_minus_alpha + daxpy_alpha + Ddot, cghip.c:969-1026).
__global__ void __launch_bounds__(BLOCK)
k_cg_fused_update(double* __restrict__ r, double* __restrict__ x,
                  const double* __restrict__ p, const double* __restrict__ t,
                  long n, const double* __restrict__ scal,
                  double* __restrict__ partials) {
    const double alpha = safe_div(scal[S_RR], scal[S_PT]);
    double acc = 0.0;
    const long stride = (long)gridDim.x * BLOCK;
    // t and x are single-use streams this iteration: non-temporal keeps
    // p (the next SpMV's gather source) and r resident in L2/L3
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
        const double rn = r[i] - alpha * ld_nt(t + i);
        r[i] = rn;
        __builtin_nontemporal_store(ld_nt(x + i) + alpha * p[i], x + i);
        acc += rn * rn;
    }
    acc = block_reduce(acc);
    if (threadIdx.x == 0) partials[blockIdx.x] = acc;
}

__device__ __forceinline__ void pipelined_coeffs(const double* scal, int first,
                                                 double* beta, double* alpha) {
    const double gamma = scal[S_GAMMA], delta = scal[S_DELTA];
    if (first) { *beta = 0.0; *alpha = safe_div(gamma, delta); }
    else {
        const double b = safe_div(gamma, scal[S_GAMMA_PREV]);
        *beta = b;
        *alpha = safe_div(gamma,
                          delta - b * safe_div(gamma, scal[S_ALPHA_PREV]));
    }
}

// pipelined-CG fused 6-vector update (Ghysels-Vanroose), device scalars,
// WITH the next iteration's dots fused in: after updating r,w this kernel
// already holds the new values in registers, so gamma' = (r',r') and
// delta' = (w',r') cost zero extra memory traffic.  The separate dot2 pass
// of the reference (two hipblasDdot, cghip.c:1735-1752) disappears from
// the iteration.
template <bool NTU>
__global__ void __launch_bounds__(BLOCK)
k_pipelined_fused(double* __restrict__ z, double* __restrict__ t,
                  double* __restrict__ p, double* __restrict__ x,
                  double* __restrict__ r, double* __restrict__ w,
                  const double* __restrict__ q, long n,
                  const double* __restrict__ scal, int first,
                  double* __restrict__ partials) {
    double beta, alpha;
    pipelined_coeffs(scal, first, &beta, &alpha);
    double g = 0.0, d = 0.0;
    const long stride = (long)gridDim.x * BLOCK;
    // NTU: z,t,p,x,q are touched once per iteration -- non-temporal so
    // their ~400 MB/iter of streams do not evict w (the next SpMV's
    // gather source) or r from L2/L3.
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
        const double zi = (NTU ? ld_nt(q + i) : q[i]) + beta * (NTU ? ld_nt(z + i) : z[i]);
        const double ti = w[i] + beta * (NTU ? ld_nt(t + i) : t[i]);
        const double pi = r[i] + beta * (NTU ? ld_nt(p + i) : p[i]);
        const double xn = (NTU ? ld_nt(x + i) : x[i]) + alpha * pi;
        if (NTU) {
            __builtin_nontemporal_store(zi, z + i);
            __builtin_nontemporal_store(ti, t + i);
            __builtin_nontemporal_store(pi, p + i);
            __builtin_nontemporal_store(xn, x + i);
        } else {
            z[i] = zi; t[i] = ti; p[i] = pi; x[i] = xn;
        }
        const double rn = r[i] - alpha * ti;
        const double wn = w[i] - alpha * zi;
        r[i] = rn; w[i] = wn;
        g += rn * rn;
        d += wn * rn;
    }
    g = block_reduce(g);
    __syncthreads();
    d = block_reduce(d);
    if (threadIdx.x == 0) {
        partials[blockIdx.x] = g;
        partials[MAXG + blockIdx.x] = d;
    }
}

// ---------------------------------------------------------------------------
// MEGAFUSED pipelined-CG iteration: SELL SpMV with the entire Ghysels-
// Vanroose 6-vector update + both next-iteration dots fused into the SpMV
// epilogue.  Requires a double-buffered w (the SpMV gathers w_old while the
// epilogue writes w_new), which makes the per-row update race-free: every
// other stream (z,t,p,x,r) is touched only at the row's own index.
// The whole iteration becomes ONE kernel (+ a 1-block finalize), and the
// intermediate q vector disappears entirely (never stored, never re-read):
// per-iteration HBM traffic drops from SELL + 15n doubles to SELL + 11n.
//
// Split-SpMV distribution support: the matA pass (MATO=false) fully updates
// interior rows (no ghost couplings) and defers border rows by storing
// their partial q into qpart; the matO pass (MATO=true) adds the ghost
// contributions, reads qpart, and updates the border rows -- so the halo
// exchange still overlaps the matA pass exactly like the reference's split
// (cghip.c:887-931).
template <typename ColT, bool NT, int UNROLL, bool MATO>
__global__ void __launch_bounds__(BLOCK)
k_sell_pipe(long nslices, long nrows_pass, long rowbase, long border_base,
            const long* __restrict__ sellptr, const ColT* __restrict__ cols,
            const double* __restrict__ vals,
            const double* __restrict__ w_old,  // gather source (nlocal)
            double* __restrict__ qpart,        // border q staging [nowned-border_base]
            double* __restrict__ z, double* __restrict__ t,
            double* __restrict__ p, double* __restrict__ x,
            double* __restrict__ r, double* __restrict__ w_new,
            const double* __restrict__ scal, int first,
            double* __restrict__ partials, long partials_off) {
    const int lane = threadIdx.x & (WAVE - 1);
    const long wslice = ((long)blockIdx.x * BLOCK + threadIdx.x) >> 6;
    const long nw = ((long)gridDim.x * BLOCK) >> 6;
    double beta, alpha;
    pipelined_coeffs(scal, first, &beta, &alpha);
    double g = 0.0, d = 0.0;
    for (long s = wslice; s < nslices; s += nw) {
        const long base = sellptr[s];
        const long len = (sellptr[s + 1] - base) >> 6;
        const double* __restrict__ v = vals + base + lane;
        const ColT* __restrict__ c = cols + base + lane;
        double sum = 0.0;
        long j = 0;
        for (; j + UNROLL <= len; j += UNROLL) {
            double a[UNROLL], xx[UNROLL];
            #pragma unroll
            for (int u = 0; u < UNROLL; ++u) {
                a[u] = NT ? ld_nt(v + (j + u) * WAVE) : v[(j + u) * WAVE];
                xx[u] = w_old[c[(j + u) * WAVE]];
            }
            #pragma unroll
            for (int u = 0; u < UNROLL; ++u) sum += a[u] * xx[u];
        }
        for (; j < len; ++j)
            sum += (NT ? ld_nt(v + j * WAVE) : v[j * WAVE]) * w_old[c[j * WAVE]];
        const long rr = s * WAVE + lane;
        if (rr < nrows_pass) {
            const long row = rowbase + rr;
            if (!MATO && row >= border_base) {
                qpart[row - border_base] = sum;  // defer to the matO pass
            } else {
                const double q = MATO ? sum + qpart[row - border_base] : sum;
                const double zi = q + beta * ld_nt(z + row);
                const double ti = w_old[row] + beta * ld_nt(t + row);
                const double pi = r[row] + beta * ld_nt(p + row);
                __builtin_nontemporal_store(zi, z + row);
                __builtin_nontemporal_store(ti, t + row);
                __builtin_nontemporal_store(pi, p + row);
                __builtin_nontemporal_store(ld_nt(x + row) + alpha * pi, x + row);
                const double rn = r[row] - alpha * ti;
                const double wn = w_old[row] - alpha * zi;
                r[row] = rn; w_new[row] = wn;
                g += rn * rn;
                d += wn * rn;
            }
        }
    }
    g = block_reduce(g);
    __syncthreads();
    d = block_reduce(d);
    if (threadIdx.x == 0) {
        partials[partials_off + blockIdx.x] = g;
        partials[MAXG + partials_off + blockIdx.x] = d;
    }
}

// one-block epilogue of a pipelined iteration: persist gamma_prev/alpha_prev
// from the OLD gamma/delta, then overwrite gamma/delta with the freshly
// reduced sums from k_pipelined_fused's partials.
__global__ void __launch_bounds__(BLOCK)
k_pipelined_finalize(const double* __restrict__ partials, int nblocks,
                     double* scal, int first) {
    double g = 0.0, d = 0.0;
    for (int i = threadIdx.x; i < nblocks; i += BLOCK) {
        g += partials[i];
        d += partials[MAXG + i];
    }
    g = block_reduce(g);
    __syncthreads();
    d = block_reduce(d);
    if (threadIdx.x == 0) {
        double beta, alpha;
        pipelined_coeffs(scal, first, &beta, &alpha);
        scal[S_GAMMA_PREV] = scal[S_GAMMA];
        scal[S_ALPHA_PREV] = alpha;
        scal[S_GAMMA] = g;
        scal[S_DELTA] = d;
    }
}

// ---------------------------------------------------------------------------
// Device-side stencil-operator generation: build the slab-local SELL arrays
// for a block-stencil SPD operator directly in HBM.  The operator is
// analytic (offsets + dof x dof blocks), so there is no reason to assemble
// 100+ GB of CSR on the host and copy it over PCIe: a 2048^3 7-pt Poisson
// slab (~90 GB of SELL data per GPU) generates in well under a second at
// HBM write speed, sized for the 288 GB of HBM3E per GPU.
// Mirrors acg_amd/gen/stencil.py::stencil_local_slab (same plane ordering:
// interior | border | ghost, ghosts sorted by (owner, global id)); column
// order within a row is enumeration order (SpMV needs no sorted columns).
//
// pb: plane z (+1 shift) -> local node base, -1 if absent  (int64[gz+2])
// offs: ksten stencil offsets as (dx, dy, dz, w) doubles
// zs_of_plane: local plane index -> z                      (int32[nplanes])
// blocks: M (offblock) then D (diagblock), dof*dof doubles each
// filter_ghost: 0 = keep cols with local node < nown_nodes (matA),
//               1 = keep ghost cols only (matO; row range given by rowbase)

__device__ __forceinline__ long stencil_col_node(
    int xi, int yi, int zi, int dx, int dy, int dz, int gx, int gy, int gz,
    const long* __restrict__ pb) {
    const int nx = xi + dx, ny = yi + dy, nz = zi + dz;
    if (nx < 0 || nx >= gx || ny < 0 || ny >= gy || nz < 0 || nz >= gz)
        return -1;
    const long base = pb[nz + 1];
    if (base < 0) return -1;
    return base + nx + (long)gx * ny;
}

__global__ void __launch_bounds__(BLOCK)
k_stencil_rowlen(long nrows_nodes, long row0_node, int gx, int gy, int gz,
                 int dof, long nown_nodes,
                 const int* __restrict__ zs_of_plane,
                 const long* __restrict__ pb,
                 const double* __restrict__ offs, int ksten,
                 int filter_ghost, long* __restrict__ rowlen) {
    const long stride = (long)gridDim.x * BLOCK;
    const long plane_nodes = (long)gx * gy;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nrows_nodes;
         i += stride) {
        const long node = row0_node + i;
        const int pl = (int)(node / plane_nodes);
        const long rem = node - (long)pl * plane_nodes;
        const int xi = (int)(rem % gx), yi = (int)(rem / gx);
        const int zi = zs_of_plane[pl];
        long cnt = filter_ghost ? 0 : 1;  // self is always owned
        for (int o = 0; o < ksten; ++o) {
            const int dx = (int)offs[o * 4 + 0], dy = (int)offs[o * 4 + 1],
                      dz = (int)offs[o * 4 + 2];
            const long cn = stencil_col_node(xi, yi, zi, dx, dy, dz, gx, gy, gz, pb);
            if (cn < 0) continue;
            const bool ghost = cn >= nown_nodes;
            if (ghost == (bool)filter_ghost) ++cnt;
        }
        rowlen[i] = cnt * dof;  // every (node,a) row has the same length
    }
}

__global__ void __launch_bounds__(BLOCK)
k_stencil_fill(long nrows_nodes, long row0_node, int gx, int gy, int gz,
               int dof, long nown_nodes,
               const int* __restrict__ zs_of_plane,
               const long* __restrict__ pb,
               const double* __restrict__ offs, int ksten,
               const double* __restrict__ blocks,  // M then D, dof*dof each
               int filter_ghost,
               const long* __restrict__ sellptr,
               int* __restrict__ cols, double* __restrict__ vals) {
    const long stride = (long)gridDim.x * BLOCK;
    const long plane_nodes = (long)gx * gy;
    const double* M = blocks;
    const double* D = blocks + dof * dof;
    const long nrows = nrows_nodes * dof;
    for (long row = (long)blockIdx.x * BLOCK + threadIdx.x; row < nrows;
         row += stride) {
        const long nodei = row / dof;
        const int a = (int)(row - nodei * dof);
        const long node = row0_node + nodei;
        const int pl = (int)(node / plane_nodes);
        const long rem = node - (long)pl * plane_nodes;
        const int xi = (int)(rem % gx), yi = (int)(rem / gx);
        const int zi = zs_of_plane[pl];
        const long s = row >> 6;
        const int lane = (int)(row & 63);
        const long base = sellptr[s];
        const long len = (sellptr[s + 1] - base) >> 6;
        long j = 0;
        if (!filter_ghost) {  // self block first
            for (int bb = 0; bb < dof; ++bb, ++j) {
                cols[base + j * WAVE + lane] = (int)(node * dof + bb);
                vals[base + j * WAVE + lane] = D[a * dof + bb];
            }
        }
        for (int o = 0; o < ksten; ++o) {
            const int dx = (int)offs[o * 4 + 0], dy = (int)offs[o * 4 + 1],
                      dz = (int)offs[o * 4 + 2];
            const double w = offs[o * 4 + 3];
            const long cn = stencil_col_node(xi, yi, zi, dx, dy, dz, gx, gy, gz, pb);
            if (cn < 0) continue;
            const bool ghost = cn >= nown_nodes;
            if (ghost != (bool)filter_ghost) continue;
            for (int bb = 0; bb < dof; ++bb, ++j) {
                cols[base + j * WAVE + lane] = (int)(cn * dof + bb);
                vals[base + j * WAVE + lane] = w * M[a * dof + bb];
            }
        }
        // pad to slice length: self column, zero value
        const int selfcol = (int)(filter_ghost ? 0 : node * dof + a);
        for (; j < len; ++j) {
            cols[base + j * WAVE + lane] = selfcol;
            vals[base + j * WAVE + lane] = 0.0;
        }
    }
}

void spmv_bsell(long nslices, long nnodes, int dof, uintptr_t bptr,
                uintptr_t bcol, uintptr_t bvals, uintptr_t x, uintptr_t y,
                uintptr_t partials, uintptr_t scal, int dotslot,
                bool dot_accum, uintptr_t stream) {
    if (nnodes == 0) return;
    long blocks = (nslices * WAVE + BLOCK - 1) / BLOCK;
    if (blocks > MAXG) blocks = MAXG;
    const bool fuse = partials != 0 && dotslot >= 0;
    dim3 g((unsigned)blocks), b(BLOCK);
    #define LBS(D, FD) \
        hipLaunchKernelGGL((k_spmv_bsell<D, FD>), g, b, 0, (hipStream_t)stream, \
            nslices, nnodes, (const long*)bptr, (const int*)bcol, \
            (const double*)bvals, (const double*)x, (double*)y, (double*)partials)
    switch (dof) {
        case 2: if (fuse) { LBS(2, true); } else { LBS(2, false); } break;
        case 3: if (fuse) { LBS(3, true); } else { LBS(3, false); } break;
        case 4: if (fuse) { LBS(4, true); } else { LBS(4, false); } break;
        default: throw std::runtime_error("spmv_bsell: dof must be 2/3/4");
    }
    #undef LBS
    check_hip("spmv_bsell");
    if (fuse) {
        hipLaunchKernelGGL(k_reduce_partials, dim3(1), dim3(BLOCK), 0,
                           (hipStream_t)stream, (const double*)partials,
                           (int)blocks, (double*)scal, dotslot, dot_accum ? 1 : 0);
        check_hip("spmv_bsell_reduce");
    }
}

void spmv_bsell_daypx(long nslices, long nnodes, int dof, uintptr_t bptr,
                      uintptr_t bcol, uintptr_t bvals, uintptr_t pold,
                      uintptr_t rvec, uintptr_t pnew, uintptr_t y,
                      uintptr_t scal, uintptr_t partials, int dotslot,
                      uintptr_t stream) {
    if (nnodes == 0) return;
    long blocks = (nslices * WAVE + BLOCK - 1) / BLOCK;
    if (blocks > MAXG) blocks = MAXG;
    dim3 g((unsigned)blocks), b(BLOCK);
    #define LBD(D) \
        hipLaunchKernelGGL((k_spmv_bsell_daypx<D>), g, b, 0, (hipStream_t)stream, \
            nslices, nnodes, (const long*)bptr, (const int*)bcol, \
            (const double*)bvals, (const double*)pold, (const double*)rvec, \
            (double*)pnew, (double*)y, (const double*)scal, (double*)partials)
    switch (dof) {
        case 2: LBD(2); break;
        case 3: LBD(3); break;
        case 4: LBD(4); break;
        default: throw std::runtime_error("spmv_bsell_daypx: dof must be 2/3/4");
    }
    #undef LBD
    check_hip("spmv_bsell_daypx");
    hipLaunchKernelGGL(k_reduce_partials, dim3(1), dim3(BLOCK), 0,
                       (hipStream_t)stream, (const double*)partials,
                       (int)blocks, (double*)scal, dotslot, 0);
    check_hip("spmv_bsell_daypx_reduce");
}

void stencil_blocklen(long nnodes, int gx, int gy, int gz, long nown_nodes,
                      uintptr_t zs_of_plane, uintptr_t pb, uintptr_t offs,
                      int ksten, uintptr_t blocklen, uintptr_t stream) {
    hipLaunchKernelGGL(k_stencil_blocklen, dim3((unsigned)elem_grid(nnodes)),
                       dim3(BLOCK), 0, (hipStream_t)stream,
                       nnodes, gx, gy, gz, nown_nodes, (const int*)zs_of_plane,
                       (const long*)pb, (const double*)offs, ksten,
                       (long*)blocklen);
    check_hip("stencil_blocklen");
}

void stencil_bfill(long nnodes, int gx, int gy, int gz, int dof,
                   long nown_nodes, uintptr_t zs_of_plane, uintptr_t pb,
                   uintptr_t offs, int ksten, uintptr_t blocks_md,
                   uintptr_t bptr, uintptr_t bcol, uintptr_t bvals,
                   uintptr_t stream) {
    dim3 g((unsigned)elem_grid(nnodes)), b(BLOCK);
    #define LBF(D) \
        hipLaunchKernelGGL((k_stencil_bfill<D>), g, b, 0, (hipStream_t)stream, \
            nnodes, gx, gy, gz, nown_nodes, (const int*)zs_of_plane, \
            (const long*)pb, (const double*)offs, ksten, \
            (const double*)blocks_md, (const long*)bptr, (int*)bcol, \
            (double*)bvals)
    switch (dof) {
        case 2: LBF(2); break;
        case 3: LBF(3); break;
        case 4: LBF(4); break;
        default: throw std::runtime_error("stencil_bfill: dof must be 2/3/4");
    }
    #undef LBF
    check_hip("stencil_bfill");
}

void stencil_rowlen(long nrows_nodes, long row0_node, int gx, int gy, int gz,
                    int dof, long nown_nodes, uintptr_t zs_of_plane,
                    uintptr_t pb, uintptr_t offs, int ksten, int filter_ghost,
                    uintptr_t rowlen, uintptr_t stream) {
    hipLaunchKernelGGL(k_stencil_rowlen, dim3((unsigned)elem_grid(nrows_nodes)),
                       dim3(BLOCK), 0, (hipStream_t)stream,
                       nrows_nodes, row0_node, gx, gy, gz, dof, nown_nodes,
                       (const int*)zs_of_plane, (const long*)pb,
                       (const double*)offs, ksten, filter_ghost, (long*)rowlen);
    check_hip("stencil_rowlen");
}

void stencil_fill(long nrows_nodes, long row0_node, int gx, int gy, int gz,
                  int dof, long nown_nodes, uintptr_t zs_of_plane,
                  uintptr_t pb, uintptr_t offs, int ksten, uintptr_t blocks,
                  int filter_ghost, uintptr_t sellptr, uintptr_t cols,
                  uintptr_t vals, uintptr_t stream) {
    const long nrows = nrows_nodes * dof;
    hipLaunchKernelGGL(k_stencil_fill, dim3((unsigned)elem_grid(nrows)),
                       dim3(BLOCK), 0, (hipStream_t)stream,
                       nrows_nodes, row0_node, gx, gy, gz, dof, nown_nodes,
                       (const int*)zs_of_plane, (const long*)pb,
                       (const double*)offs, ksten, (const double*)blocks,
                       filter_ghost, (const long*)sellptr, (int*)cols,
                       (double*)vals);
    check_hip("stencil_fill");
}

// ---------------------------------------------------------------------------
// MATRIX-FREE stencil operator (dof=1, constant coefficients -- 5/7/27-pt
// Poisson-type).  The assembled operator re-reads 12 B/nnz of vals+cols per
// SpMV that the stencil makes redundant: y[i] = diag*x[i] + sum w*x[nb] is
// fully determined by the grid coordinates.  Applying it matrix-free reads
// ONLY x (largely L2-cached across the 7/27 neighbour touches) and writes y,
// dropping per-iteration HBM traffic by the whole vals+cols stream -- and
// removes the operator from HBM entirely (a 2048^3 7-pt slab is ~90 GB of
// SELL; matrix-free it is 0).  Beyond the reference (aCG always assembles);
// opt-in because real matrices (Queen_4147) have position-dependent values
// and MUST be measured with the memory-resident operator.
//
// Same matA/matO split as the assembled path: the matA pass covers owned
// couplings (diag included), the matO pass adds ghost-plane couplings to
// border rows after the halo lands.
template <bool MATO, bool FUSE_DOT>
__global__ void __launch_bounds__(BLOCK)
k_stencil_spmv(long nrows_nodes, long row0_node, int gx, int gy, int gz,
               long nown_nodes,
               const int* __restrict__ zs_of_plane,
               const long* __restrict__ pb,
               const double* __restrict__ offs, int ksten, double diag,
               const double* __restrict__ x, double* __restrict__ y,
               double* __restrict__ partials) {
    const long stride = (long)gridDim.x * BLOCK;
    const long plane_nodes = (long)gx * gy;
    double dacc = 0.0;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nrows_nodes;
         i += stride) {
        const long node = row0_node + i;
        const int pl = (int)(node / plane_nodes);
        const long rem = node - (long)pl * plane_nodes;
        const int xi = (int)(rem % gx), yi = (int)(rem / gx);
        const int zi = zs_of_plane[pl];
        double sum = MATO ? 0.0 : diag * x[node];
        for (int o = 0; o < ksten; ++o) {
            const int dx = (int)offs[o * 4 + 0], dy = (int)offs[o * 4 + 1],
                      dz = (int)offs[o * 4 + 2];
            const long cn = stencil_col_node(xi, yi, zi, dx, dy, dz,
                                             gx, gy, gz, pb);
            if (cn < 0) continue;
            if ((cn >= nown_nodes) == MATO) sum += offs[o * 4 + 3] * x[cn];
        }
        if (MATO) y[node] += sum; else y[node] = sum;
        if (FUSE_DOT) dacc += x[node] * sum;
    }
    if (FUSE_DOT) {
        dacc = block_reduce(dacc);
        if (threadIdx.x == 0) partials[blockIdx.x] = dacc;
    }
}

// Megafused matrix-free pipelined iteration: identical epilogue and qpart /
// partials protocol to k_sell_pipe (see there), with the SpMV computed from
// the stencil instead of SELL loads.  Per-iteration traffic collapses to
// the 11n-double vector stream alone.
template <bool MATO>
__global__ void __launch_bounds__(BLOCK)
k_stencil_pipe(long nrows_nodes, long row0_node, long border_base,
               int gx, int gy, int gz, long nown_nodes,
               const int* __restrict__ zs_of_plane,
               const long* __restrict__ pb,
               const double* __restrict__ offs, int ksten, double diag,
               const double* __restrict__ w_old, double* __restrict__ qpart,
               double* __restrict__ z, double* __restrict__ t,
               double* __restrict__ p, double* __restrict__ xv,
               double* __restrict__ r, double* __restrict__ w_new,
               const double* __restrict__ scal, int first,
               double* __restrict__ partials, long partials_off) {
    const long stride = (long)gridDim.x * BLOCK;
    const long plane_nodes = (long)gx * gy;
    double beta, alpha;
    pipelined_coeffs(scal, first, &beta, &alpha);
    double g = 0.0, d = 0.0;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nrows_nodes;
         i += stride) {
        const long node = row0_node + i;
        const int pl = (int)(node / plane_nodes);
        const long rem = node - (long)pl * plane_nodes;
        const int xi = (int)(rem % gx), yi = (int)(rem / gx);
        const int zi = zs_of_plane[pl];
        double sum = MATO ? 0.0 : diag * w_old[node];
        for (int o = 0; o < ksten; ++o) {
            const int dx = (int)offs[o * 4 + 0], dy = (int)offs[o * 4 + 1],
                      dz = (int)offs[o * 4 + 2];
            const long cn = stencil_col_node(xi, yi, zi, dx, dy, dz,
                                             gx, gy, gz, pb);
            if (cn < 0) continue;
            if ((cn >= nown_nodes) == MATO) sum += offs[o * 4 + 3] * w_old[cn];
        }
        const long row = node;  // dof = 1
        if (!MATO && row >= border_base) {
            qpart[row - border_base] = sum;  // defer to the matO pass
        } else {
            const double q = MATO ? sum + qpart[row - border_base] : sum;
            const double zi_ = q + beta * ld_nt(z + row);
            const double ti = w_old[row] + beta * ld_nt(t + row);
            const double pi = r[row] + beta * ld_nt(p + row);
            __builtin_nontemporal_store(zi_, z + row);
            __builtin_nontemporal_store(ti, t + row);
            __builtin_nontemporal_store(pi, p + row);
            __builtin_nontemporal_store(ld_nt(xv + row) + alpha * pi, xv + row);
            const double rn = r[row] - alpha * ti;
            const double wn = w_old[row] - alpha * zi_;
            r[row] = rn; w_new[row] = wn;
            g += rn * rn;
            d += wn * rn;
        }
    }
    g = block_reduce(g);
    __syncthreads();
    d = block_reduce(d);
    if (threadIdx.x == 0) {
        partials[partials_off + blockIdx.x] = g;
        partials[MAXG + partials_off + blockIdx.x] = d;
    }
}

// 7-pt z-column-walk specialisation: one thread owns one (x,y) column and
// walks the owned planes z0..z1-1, carrying the z-1/z/z+1 values of w_old
// in registers.  The generic kernel re-reads the z +- 1 planes through an
// L2 that cannot hold them (a 512^2 plane is 2 MB x thousands of in-flight
// blocks), costing ~2 extra w_old streams; the walk reads w_old EXACTLY
// once.  x +- 1 / y +- gx neighbours come from L1 (adjacent lanes load
// them).  matA pass only -- ghost-plane couplings stay with the generic
// matO kernel (2 planes at most).
template <bool FUSE_DOT>
__global__ void __launch_bounds__(BLOCK)
k_stencil_spmv7(long plane_nodes, int gx, int z0, int z1, int gz,
                long nown_nodes, const long* __restrict__ pb,
                double diag, double wxm, double wxp, double wym, double wyp,
                double wzm, double wzp,
                const double* __restrict__ x, double* __restrict__ y,
                double* __restrict__ partials) {
    const long stride = (long)gridDim.x * BLOCK;
    double dacc = 0.0;
    for (long xy = (long)blockIdx.x * BLOCK + threadIdx.x; xy < plane_nodes;
         xy += stride) {
        const int xi = (int)(xy % gx);
        const long yi = xy / gx;
        const bool hxm = xi > 0, hxp = xi < gx - 1;
        const bool hym = yi > 0, hyp = yi < plane_nodes / gx - 1;
        long base_c = pb[z0 + 1];
        double xc = x[base_c + xy];
        const long base_m = (z0 > 0) ? pb[z0] : -1;
        bool hm = base_m >= 0 && base_m < nown_nodes;
        double xm = hm ? x[base_m + xy] : 0.0;
        for (int zz = z0; zz < z1; ++zz) {
            const long node = base_c + xy;
            const long base_p = (zz + 1 < gz) ? pb[zz + 2] : -1;
            const bool hp = base_p >= 0 && base_p < nown_nodes;
            const double xp = hp ? x[base_p + xy] : 0.0;
            double sum = diag * xc;
            if (hxm) sum += wxm * x[node - 1];
            if (hxp) sum += wxp * x[node + 1];
            if (hym) sum += wym * x[node - gx];
            if (hyp) sum += wyp * x[node + gx];
            if (hm) sum += wzm * xm;
            if (hp) sum += wzp * xp;
            y[node] = sum;
            if (FUSE_DOT) dacc += xc * sum;
            xm = xc; hm = true;
            xc = xp; base_c = base_p;
        }
    }
    if (FUSE_DOT) {
        dacc = block_reduce(dacc);
        if (threadIdx.x == 0) partials[blockIdx.x] = dacc;
    }
}

// megafused 7-pt column walk: k_stencil_pipe's epilogue on the walk above.
__global__ void __launch_bounds__(BLOCK)
k_stencil_pipe7(long plane_nodes, int gx, int z0, int z1, int gz,
                long border_base, long nown_nodes,
                const long* __restrict__ pb,
                double diag, double wxm, double wxp, double wym, double wyp,
                double wzm, double wzp,
                const double* __restrict__ w_old, double* __restrict__ qpart,
                double* __restrict__ z, double* __restrict__ t,
                double* __restrict__ p, double* __restrict__ xv,
                double* __restrict__ r, double* __restrict__ w_new,
                const double* __restrict__ scal, int first,
                double* __restrict__ partials, long partials_off) {
    const long stride = (long)gridDim.x * BLOCK;
    double beta, alpha;
    pipelined_coeffs(scal, first, &beta, &alpha);
    double g = 0.0, d = 0.0;
    for (long xy = (long)blockIdx.x * BLOCK + threadIdx.x; xy < plane_nodes;
         xy += stride) {
        const int xi = (int)(xy % gx);
        const long yi = xy / gx;
        const bool hxm = xi > 0, hxp = xi < gx - 1;
        const bool hym = yi > 0, hyp = yi < plane_nodes / gx - 1;
        long base_c = pb[z0 + 1];
        double xc = w_old[base_c + xy];
        const long base_m = (z0 > 0) ? pb[z0] : -1;
        bool hm = base_m >= 0 && base_m < nown_nodes;
        double xm = hm ? w_old[base_m + xy] : 0.0;
        for (int zz = z0; zz < z1; ++zz) {
            const long node = base_c + xy;
            const long base_p = (zz + 1 < gz) ? pb[zz + 2] : -1;
            const bool hp = base_p >= 0 && base_p < nown_nodes;
            const double xp = hp ? w_old[base_p + xy] : 0.0;
            double sum = diag * xc;
            if (hxm) sum += wxm * w_old[node - 1];
            if (hxp) sum += wxp * w_old[node + 1];
            if (hym) sum += wym * w_old[node - gx];
            if (hyp) sum += wyp * w_old[node + gx];
            if (hm) sum += wzm * xm;
            if (hp) sum += wzp * xp;
            if (node >= border_base) {
                qpart[node - border_base] = sum;  // matO finishes border rows
            } else {
                const double zi_ = sum + beta * ld_nt(z + node);
                const double ti = xc + beta * ld_nt(t + node);
                const double pi = r[node] + beta * ld_nt(p + node);
                __builtin_nontemporal_store(zi_, z + node);
                __builtin_nontemporal_store(ti, t + node);
                __builtin_nontemporal_store(pi, p + node);
                __builtin_nontemporal_store(ld_nt(xv + node) + alpha * pi,
                                            xv + node);
                const double rn = r[node] - alpha * ti;
                const double wn = xc - alpha * zi_;
                r[node] = rn; w_new[node] = wn;
                g += rn * rn;
                d += wn * rn;
            }
            xm = xc; hm = true;
            xc = xp; base_c = base_p;
        }
    }
    g = block_reduce(g);
    __syncthreads();
    d = block_reduce(d);
    if (threadIdx.x == 0) {
        partials[partials_off + blockIdx.x] = g;
        partials[MAXG + partials_off + blockIdx.x] = d;
    }
}

void stencil_spmv7(long plane_nodes, int gx, int z0, int z1, int gz,
                   long nown_nodes, uintptr_t pb, double diag,
                   double wxm, double wxp, double wym, double wyp,
                   double wzm, double wzp, uintptr_t x, uintptr_t y,
                   uintptr_t partials, uintptr_t scal, int dotslot,
                   bool dot_accum, uintptr_t stream) {
    if (plane_nodes == 0 || z1 <= z0) return;
    long blocks = (plane_nodes + BLOCK - 1) / BLOCK;
    if (blocks > MAXG) blocks = MAXG;
    const bool fuse = partials != 0 && dotslot >= 0;
    dim3 g((unsigned)blocks), b(BLOCK);
    #define LS7(FD) \
        hipLaunchKernelGGL((k_stencil_spmv7<FD>), g, b, 0, (hipStream_t)stream, \
            plane_nodes, gx, z0, z1, gz, nown_nodes, (const long*)pb, diag, \
            wxm, wxp, wym, wyp, wzm, wzp, (const double*)x, (double*)y, \
            (double*)partials)
    if (fuse) { LS7(true); } else { LS7(false); }
    #undef LS7
    check_hip("stencil_spmv7");
    if (fuse) {
        hipLaunchKernelGGL(k_reduce_partials, dim3(1), dim3(BLOCK), 0,
                           (hipStream_t)stream, (const double*)partials,
                           (int)blocks, (double*)scal, dotslot, dot_accum ? 1 : 0);
        check_hip("stencil_spmv7_reduce");
    }
}

long stencil_pipe7(long plane_nodes, int gx, int z0, int z1, int gz,
                   long border_base, long nown_nodes, uintptr_t pb,
                   double diag, double wxm, double wxp, double wym,
                   double wyp, double wzm, double wzp, uintptr_t w_old,
                   uintptr_t qpart, uintptr_t z, uintptr_t t, uintptr_t p,
                   uintptr_t xv, uintptr_t r, uintptr_t w_new, uintptr_t scal,
                   int first, uintptr_t partials, long partials_off,
                   uintptr_t stream) {
    if (plane_nodes == 0 || z1 <= z0) return 0;
    long blocks = (plane_nodes + BLOCK - 1) / BLOCK;
    const long cap = (partials_off == 0) ? (MAXG - 1024) : (MAXG - partials_off);
    if (blocks > cap) blocks = cap;  // see sell_pipe
    dim3 g((unsigned)blocks), b(BLOCK);
    hipLaunchKernelGGL(k_stencil_pipe7, g, b, 0, (hipStream_t)stream,
                       plane_nodes, gx, z0, z1, gz, border_base, nown_nodes,
                       (const long*)pb, diag, wxm, wxp, wym, wyp, wzm, wzp,
                       (const double*)w_old, (double*)qpart, (double*)z,
                       (double*)t, (double*)p, (double*)xv, (double*)r,
                       (double*)w_new, (const double*)scal, first,
                       (double*)partials, partials_off);
    check_hip("stencil_pipe7");
    return blocks;
}

void stencil_spmv(long nrows_nodes, long row0_node, int gx, int gy, int gz,
                  long nown_nodes, uintptr_t zs_of_plane, uintptr_t pb,
                  uintptr_t offs, int ksten, double diag, uintptr_t x,
                  uintptr_t y, bool mato, uintptr_t partials, uintptr_t scal,
                  int dotslot, bool dot_accum, uintptr_t stream) {
    if (nrows_nodes == 0) return;
    const long blocks = elem_grid(nrows_nodes);
    const bool fuse = partials != 0 && dotslot >= 0;
    dim3 g((unsigned)blocks), b(BLOCK);
    #define LMS(MATO, FD) \
        hipLaunchKernelGGL((k_stencil_spmv<MATO, FD>), g, b, 0, \
            (hipStream_t)stream, nrows_nodes, row0_node, gx, gy, gz, \
            nown_nodes, (const int*)zs_of_plane, (const long*)pb, \
            (const double*)offs, ksten, diag, (const double*)x, (double*)y, \
            (double*)partials)
    if (mato) { if (fuse) { LMS(true, true); } else { LMS(true, false); } }
    else      { if (fuse) { LMS(false, true); } else { LMS(false, false); } }
    #undef LMS
    check_hip("stencil_spmv");
    if (fuse) {
        hipLaunchKernelGGL(k_reduce_partials, dim3(1), dim3(BLOCK), 0,
                           (hipStream_t)stream, (const double*)partials,
                           (int)blocks, (double*)scal, dotslot, dot_accum ? 1 : 0);
        check_hip("stencil_spmv_reduce");
    }
}

long stencil_pipe(long nrows_nodes, long row0_node, long border_base,
                  int gx, int gy, int gz, long nown_nodes,
                  uintptr_t zs_of_plane, uintptr_t pb, uintptr_t offs,
                  int ksten, double diag, uintptr_t w_old, uintptr_t qpart,
                  uintptr_t z, uintptr_t t, uintptr_t p, uintptr_t xv,
                  uintptr_t r, uintptr_t w_new, uintptr_t scal, int first,
                  uintptr_t partials, long partials_off, bool mato,
                  uintptr_t stream) {
    if (nrows_nodes == 0) return 0;
    long blocks = elem_grid(nrows_nodes);
    const long cap = (partials_off == 0) ? (MAXG - 1024) : (MAXG - partials_off);
    if (blocks > cap) blocks = cap;  // see sell_pipe
    dim3 g((unsigned)blocks), b(BLOCK);
    #define LMP(MATO) \
        hipLaunchKernelGGL((k_stencil_pipe<MATO>), g, b, 0, (hipStream_t)stream, \
            nrows_nodes, row0_node, border_base, gx, gy, gz, nown_nodes, \
            (const int*)zs_of_plane, (const long*)pb, (const double*)offs, \
            ksten, diag, (const double*)w_old, (double*)qpart, (double*)z, \
            (double*)t, (double*)p, (double*)xv, (double*)r, (double*)w_new, \
            (const double*)scal, first, (double*)partials, partials_off)
    if (mato) { LMP(true); } else { LMP(false); }
    #undef LMP
    check_hip("stencil_pipe");
    return blocks;
}

// ---------------------------------------------------------------------------
// Monolithic device-side CG: the ENTIRE solver loop in one cooperative
// launch -- zero per-iteration launch/sync overhead (reference
// acgsolverhip_cg_kernel, cg-kernels-hip.hip:1386-1747; single-GPU, like
// the reference's HIP build).  MI355X design: SELL-C-64 SpMV, grid-wide
// barriers via cooperative groups, per-block partials reduced by block 0
// (deterministic).  Residency: the launcher sizes the grid from the
// occupancy query, everything inside is grid-stride.
#include <hip/hip_cooperative_groups.h>

#define RLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT
typedef __attribute__((address_space(1))) unsigned int gu32;

// Hand-rolled grid barrier (guide §6 G16): cooperative_groups::grid.sync()
// measured ~150 us per sync on ROCm 7.2.  v2 design, after measuring a
// per-block release-fence + single-word-poll version still slow at >1000
// blocks:
//  - ALL shared payloads (vectors, partials) are written WRITE-THROUGH
//    (sc1: relaxed agent-scope 8-byte atomic stores) so no release fence
//    (no per-block buffer_wbl2 storm) is needed -- every wave just drains
//    vmcnt before its block arrives (guide R1 + pitfall 14).
//  - waiters poll a PER-XCD copy of the generation word (the dispatcher
//    places block b on XCD b%8, so each poll line serves ~1/8 of the
//    blocks; correctness does not depend on the placement, every copy is
//    written).  Polling is RELAXED with s_sleep backoff; ONE acquire
//    (L1 invalidate) per block after wake.  Spins are bounded: on timeout
//    fail[0] is set and every block exits (no hang).
// barrier_state layout (u32 words): [0]=cnt(even phases), [8]=cnt(odd
// phases) (sense-reversing parity counters: the releaser resets the
// just-used counter, which no block touches again until two barriers
// later -- a racing early arrival for the NEXT barrier targets the OTHER
// word, so the reset can never eat an arrival), [1]=fail,
// [16+16*x]=generation copy for XCD x (64 B stride: distinct lines).
#define BAR_GEN0 16
#define BAR_GENSTRIDE 16
// hierarchical-barrier extension words (v3): per-XCD arrival counters on
// distinct cachelines, one set per parity: [144+16x] even, [272+16x] odd.
#define BAR_XCNT0 144
#define BAR_XCNT_PAR 128
#define BAR_STATE_WORDS 400

__device__ __forceinline__ void st_sc1_f64(double* p, double v) {
    __hip_atomic_store(reinterpret_cast<unsigned long long*>(p),
                       (unsigned long long)__double_as_longlong(v), RLX_AGENT);
}

// Hierarchical barrier (v3): arrivals first bump a PER-XCD counter line
// (blocks land on XCD b%8, so the ~128 RMWs per line proceed on 8 lines
// in parallel instead of 1024 serialised on one line -- the ~35 us cost
// of the flat barrier at 1024 blocks); each XCD's last arriver bumps the
// global counter (<= 8 RMWs); the global last resets BOTH levels of the
// current parity and bumps every generation copy.  Parity safety is the
// same argument as the flat barrier: the words being reset belong to the
// phase that no block will touch again until two barriers later.
__device__ __forceinline__ bool grid_barrier_hier(unsigned* state) {
    gu32* fail = (gu32*)(state + 1);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    bool ok = true;
    if (threadIdx.x == 0) {
        const int myxcd = (int)(blockIdx.x & 7);
        const unsigned nb = gridDim.x;
        const unsigned q = nb >> 3, rmd = nb & 7u;
        const unsigned blocks_on_mine = q + ((unsigned)myxcd < rmd ? 1u : 0u);
        const unsigned nxcd = nb < 8u ? nb : 8u;
        gu32* mygen = (gu32*)(state + BAR_GEN0 + BAR_GENSTRIDE * myxcd);
        const unsigned g = __hip_atomic_load(mygen, RLX_AGENT);
        const unsigned par = g & 1u;
        gu32* xcnt = (gu32*)(state + BAR_XCNT0 + BAR_XCNT_PAR * par
                             + BAR_GENSTRIDE * myxcd);
        const unsigned ax = __hip_atomic_fetch_add(xcnt, 1u, RLX_AGENT) + 1u;
        bool releaser = false;
        if (ax == blocks_on_mine) {
            gu32* gcnt = (gu32*)(state + (par ? 8 : 0));
            const unsigned ag = __hip_atomic_fetch_add(gcnt, 1u, RLX_AGENT) + 1u;
            if (ag == nxcd) {
                __hip_atomic_store(gcnt, 0u, RLX_AGENT);
                #pragma unroll
                for (int xx = 0; xx < 8; ++xx)
                    __hip_atomic_store(
                        (gu32*)(state + BAR_XCNT0 + BAR_XCNT_PAR * par
                                + BAR_GENSTRIDE * xx), 0u, RLX_AGENT);
                #pragma unroll
                for (int xx = 0; xx < 8; ++xx)
                    __hip_atomic_store(
                        (gu32*)(state + BAR_GEN0 + BAR_GENSTRIDE * xx),
                        g + 1u, RLX_AGENT);
                releaser = true;
            }
        }
        if (!releaser) {
            unsigned spins = 0;
            while (__hip_atomic_load(mygen, RLX_AGENT) == g) {
                if (spins < 32) __builtin_amdgcn_s_sleep(2);
                else __builtin_amdgcn_s_sleep(64);
                if (++spins > 4000000u) {
                    __hip_atomic_store(fail, 1u, RLX_AGENT);
                    ok = false;
                    break;
                }
                if (__hip_atomic_load(fail, RLX_AGENT)) { ok = false; break; }
            }
        }
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    }
    __syncthreads();
    return ok;
}

__device__ __forceinline__ bool grid_barrier(unsigned* state) {
    gu32* fail = (gu32*)(state + 1);
    // every wave: drain pending sc1 payload stores before signalling (also
    // completes this block's own reset/gen stores from the previous
    // barrier before its new arrival -- the parity-counter safety hinges
    // on this drain)
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    bool ok = true;
    if (threadIdx.x == 0) {
        const int myxcd = (int)(blockIdx.x & 7);
        gu32* mygen = (gu32*)(state + BAR_GEN0 + BAR_GENSTRIDE * myxcd);
        const unsigned g = __hip_atomic_load(mygen, RLX_AGENT);
        gu32* cnt = (gu32*)(state + ((g & 1u) ? 8 : 0));
        const unsigned arrived = __hip_atomic_fetch_add(cnt, 1u, RLX_AGENT) + 1u;
        if (arrived == gridDim.x) {
            __hip_atomic_store(cnt, 0u, RLX_AGENT);
            #pragma unroll
            for (int xx = 0; xx < 8; ++xx)
                __hip_atomic_store((gu32*)(state + BAR_GEN0 + BAR_GENSTRIDE * xx),
                                   g + 1u, RLX_AGENT);
        } else {
            unsigned spins = 0;
            while (__hip_atomic_load(mygen, RLX_AGENT) == g) {
                if (spins < 32) __builtin_amdgcn_s_sleep(2);
                else __builtin_amdgcn_s_sleep(64);
                if (++spins > 4000000u) {  // ~7 s worst case: fail fast, never hang
                    __hip_atomic_store(fail, 1u, RLX_AGENT);
                    ok = false;
                    break;
                }
                if (__hip_atomic_load(fail, RLX_AGENT)) { ok = false; break; }
            }
        }
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    }
    __syncthreads();
    return ok;
}

// results: scal[S_BNRM2]=(b,b), scal[S_RR_PREV]=(r0,r0), scal[S_RR]=final
// (r,r); out2[0]=niterations, out2[1]=converged (-1 = barrier timeout).
// barrier_state: 3 zeroed u32 words {cnt, gen, fail}.
template <typename ColT>
__global__ void __launch_bounds__(BLOCK)
k_cg_device(long nslices, long nrows,
            const long* __restrict__ sellptr, const ColT* __restrict__ cols,
            const double* __restrict__ vals,
            const double* __restrict__ b, double* __restrict__ x,
            double* __restrict__ r, double* __restrict__ p,
            double* __restrict__ t,
            double* __restrict__ scal, double* __restrict__ partials,
            int* __restrict__ out2, unsigned* __restrict__ barrier_state,
            int maxits, double res_atol, double res_rtol, int hier) {
    const long tid = (long)blockIdx.x * BLOCK + threadIdx.x;
    const long nth = (long)gridDim.x * BLOCK;
    const int lane = threadIdx.x & (WAVE - 1);
    const long wslice0 = tid >> 6;
    const long nw = nth >> 6;
    bool alive = true;
    auto gsync = [&]() {
        if (alive && !(hier ? grid_barrier_hier(barrier_state)
                            : grid_barrier(barrier_state))) alive = false;
        return alive;
    };

    // grid-wide sum helper: every block contributes partials[bid] (sc1);
    // block 0 reduces into scal[slot]; two grid barriers bracket it.
    auto grid_sum = [&](double v, int slot) {
        v = block_reduce(v);
        if (threadIdx.x == 0) st_sc1_f64(partials + blockIdx.x, v);
        if (!gsync()) return;
        if (blockIdx.x == 0) {
            double s = 0.0;
            for (int i = threadIdx.x; i < (int)gridDim.x; i += BLOCK)
                s += partials[i];
            s = block_reduce(s);
            if (threadIdx.x == 0) st_sc1_f64(scal + slot, s);
        }
        gsync();
    };
    auto spmv = [&](const double* __restrict__ xin, double* __restrict__ yout,
                    bool fuse) -> double {
        double dacc = 0.0;
        for (long s = wslice0; s < nslices; s += nw) {
            const long base = sellptr[s];
            const long len = (sellptr[s + 1] - base) >> 6;
            const double* __restrict__ v = vals + base + lane;
            const ColT* __restrict__ c = cols + base + lane;
            double sum = 0.0;
            long j = 0;
            for (; j + 4 <= len; j += 4) {
                double a0 = ld_nt(v + (j + 0) * WAVE), x0 = xin[c[(j + 0) * WAVE]];
                double a1 = ld_nt(v + (j + 1) * WAVE), x1 = xin[c[(j + 1) * WAVE]];
                double a2 = ld_nt(v + (j + 2) * WAVE), x2 = xin[c[(j + 2) * WAVE]];
                double a3 = ld_nt(v + (j + 3) * WAVE), x3 = xin[c[(j + 3) * WAVE]];
                sum += a0 * x0; sum += a1 * x1; sum += a2 * x2; sum += a3 * x3;
            }
            for (; j < len; ++j) sum += ld_nt(v + j * WAVE) * xin[c[j * WAVE]];
            const long row = s * WAVE + lane;
            if (row < nrows) {
                st_sc1_f64(yout + row, sum);
                if (fuse) dacc += xin[row] * sum;
            }
        }
        return dacc;
    };

    // bnrm2; r0 = b - A x0; p = r0; rr0
    double acc = 0.0;
    for (long i = tid; i < nrows; i += nth) acc += b[i] * b[i];
    grid_sum(acc, S_BNRM2);
    spmv(x, t, false);
    gsync();
    acc = 0.0;
    for (long i = tid; i < nrows; i += nth) {
        const double ri = b[i] - t[i];
        st_sc1_f64(r + i, ri);
        st_sc1_f64(p + i, ri);
        acc += ri * ri;
    }
    grid_sum(acc, S_RR);
    const double bnrm2sqr = alive ? scal[S_BNRM2] : 1.0;
    double rr = alive ? scal[S_RR] : 0.0;
    if (tid == 0) st_sc1_f64(scal + S_RR_PREV, rr);  // report (r0,r0)
    const double rt = res_rtol * sqrt(bnrm2sqr) > res_atol
        ? res_rtol * sqrt(bnrm2sqr) : res_atol;
    const double rtol2 = rt * rt;
    int k = 0, converged = (rtol2 > 0.0 && rr <= rtol2) ? 1 : 0;
    while (alive && !converged && k < maxits) {
        // t = A p (fused (p,t))
        gsync();  // p is consistent (written by all blocks last iter)
        const double pt_part = spmv(p, t, true);
        grid_sum(pt_part, S_PT);
        if (!alive) break;
        const double alpha = safe_div(rr, scal[S_PT]);
        acc = 0.0;
        for (long i = tid; i < nrows; i += nth) {
            const double rn = r[i] - alpha * t[i];
            st_sc1_f64(r + i, rn);
            st_sc1_f64(x + i, x[i] + alpha * p[i]);
            acc += rn * rn;
        }
        grid_sum(acc, S_RR);
        if (!alive) break;
        const double rr_new = scal[S_RR];
        const double beta = rr_new / rr;
        for (long i = tid; i < nrows; i += nth)
            st_sc1_f64(p + i, beta * p[i] + r[i]);
        rr = rr_new;
        ++k;
        if (rtol2 > 0.0 && rr <= rtol2) converged = 1;
    }
    if (tid == 0) {
        st_sc1_f64(scal + S_RR, rr);
        out2[0] = k;
        out2[1] = alive ? converged : -1;
    }
    (void)bnrm2sqr;
}

// fp64 atomic-scatter throughput probe (design study for a
// symmetric-storage SpMV: the transpose half's y updates would be HW
// atomic adds; whether that can beat the full-storage roofline depends
// entirely on sustained global_atomic_add_f64 throughput at realistic
// target distributions).  MODE 0: atomic add; 1: plain store (upper
// bound); 2: gather-read (calibration).
__global__ void __launch_bounds__(BLOCK)
k_atomic_probe(double* __restrict__ y, const int* __restrict__ idx,
               const double* __restrict__ v, long nnz, int mode) {
    const long stride = (long)gridDim.x * BLOCK;
    double acc = 0.0;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < nnz; i += stride) {
        if (mode == 0) unsafeAtomicAdd(&y[idx[i]], v[i]);
        else if (mode == 1) y[idx[i]] = v[i];
        else acc += y[idx[i]] * v[i];
    }
    if (mode == 2 && acc == -1.0) y[0] = acc;  // keep the reads live
}

// halo pack: sendbuf[i] = x[sendidx[i]]
// (reference acghalo_pack_hip_double, halo-kernels-hip.hip:48-103; no unpack
// kernel exists -- ghosts are received in place, see dist/halo.py)
template <typename IdxT>
__global__ void __launch_bounds__(BLOCK)
k_pack_gather(double* __restrict__ sendbuf, const double* __restrict__ x,
              const IdxT* __restrict__ idx, long n) {
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride)
        sendbuf[i] = x[idx[i]];
}

// ---------------------------------------------------------------------------
// host-side launchers (pybind).  Tensors arrive as raw device pointers +
// sizes + the caller's HIP stream handle (torch.cuda.current_stream().cuda_stream);
// no torch C++ dependency, no hipify, plain HIP throughout.

using std::uintptr_t;

static inline hipStream_t S(uintptr_t s) { return (hipStream_t)s; }

static void reduce_partials(uintptr_t partials, int nblocks, uintptr_t scal,
                            int slot, bool accumulate, uintptr_t stream) {
    hipLaunchKernelGGL(k_reduce_partials, dim3(1), dim3(BLOCK), 0, S(stream),
                       (const double*)partials, nblocks, (double*)scal, slot,
                       accumulate ? 1 : 0);
    check_hip("reduce_partials");
}

void spmv(long nrows, long rowbase, uintptr_t rowptr, uintptr_t colidx,
          int col64, uintptr_t vals, uintptr_t x, uintptr_t y,
          int lanes, bool accum, uintptr_t partials, uintptr_t scal,
          int dotslot, bool dot_accum, uintptr_t stream) {
    if (nrows == 0) return;
    const int rows_per_block = BLOCK / lanes;
    long blocks = (nrows + rows_per_block - 1) / rows_per_block;
    if (blocks > MAXG) blocks = MAXG;
    const bool fuse = partials != 0 && dotslot >= 0;
    dim3 g((unsigned)blocks), b(BLOCK);
    #define LAUNCH_SPMV(CT, L, AC, FD) \
        hipLaunchKernelGGL((spmv_csr_vector<CT, L, AC, FD>), g, b, 0, S(stream), \
            nrows, rowbase, (const long*)rowptr, (const CT*)colidx, \
            (const double*)vals, (const double*)x, (double*)y, (double*)partials)
    #define DISPATCH_L(CT, AC, FD) \
        switch (lanes) { \
            case 4:  LAUNCH_SPMV(CT, 4,  AC, FD); break; \
            case 8:  LAUNCH_SPMV(CT, 8,  AC, FD); break; \
            case 16: LAUNCH_SPMV(CT, 16, AC, FD); break; \
            case 32: LAUNCH_SPMV(CT, 32, AC, FD); break; \
            case 64: LAUNCH_SPMV(CT, 64, AC, FD); break; \
            default: throw std::runtime_error("spmv: lanes must be 4/8/16/32/64"); }
    #define DISPATCH_AC(CT) \
        if (accum) { if (fuse) { DISPATCH_L(CT, true, true) } else { DISPATCH_L(CT, true, false) } } \
        else       { if (fuse) { DISPATCH_L(CT, false, true) } else { DISPATCH_L(CT, false, false) } }
    if (col64) { DISPATCH_AC(long) } else { DISPATCH_AC(int) }
    #undef DISPATCH_AC
    #undef DISPATCH_L
    #undef LAUNCH_SPMV
    check_hip("spmv");
    if (fuse)
        reduce_partials(partials, (int)blocks, scal, dotslot, dot_accum, stream);
}

// Row-binned hybrid CSR SpMV: rows pre-sorted by length into bins, one
// launch per bin with bin-appropriate LANES (4..64).  Every row is fully
// reduced by one lane group => deterministic, no atomics; each launch
// writes its own partials window, one finalize over the union.  This is
// the MI355X-native load-balancer for power-law rows (reference analog:
// merge-path csrgemv_merge, cg-kernels-hip.hip:348-1175).
// bins: (start_in_rowlist, count, lanes) triples.
void spmv_binned(long rowbase, uintptr_t rowptr, uintptr_t colidx, int col64,
                 uintptr_t vals, uintptr_t x, uintptr_t y, uintptr_t rowlist,
                 const std::vector<std::array<long, 3>>& bins,
                 bool accum, uintptr_t partials, uintptr_t scal,
                 int dotslot, bool dot_accum, uintptr_t stream) {
    const bool fuse = partials != 0 && dotslot >= 0;
    long poff = 0;
    for (const auto& bin : bins) {
        const long start = bin[0], count = bin[1];
        const int lanes = (int)bin[2];
        if (count == 0) continue;
        const int rows_per_block = BLOCK / lanes;
        long blocks = (count + rows_per_block - 1) / rows_per_block;
        if (blocks > 3072) blocks = 3072;  // grid-stride within the bin
        // (5 bins x 3072 <= MAXG partials)
        if (poff + blocks > MAXG)
            throw std::runtime_error("spmv_binned: partials overflow");
        dim3 g((unsigned)blocks), b(BLOCK);
        const int* rl = (const int*)rowlist + start;
        double* pp = (double*)partials + poff;
        #define LAUNCH_BIN(CT, L, AC, FD) \
            hipLaunchKernelGGL((spmv_csr_vector<CT, L, AC, FD, true>), g, b, 0, \
                S(stream), count, rowbase, (const long*)rowptr, \
                (const CT*)colidx, (const double*)vals, (const double*)x, \
                (double*)y, pp, rl)
        #define DISPATCH_BL(CT, AC, FD) \
            switch (lanes) { \
                case 4:  LAUNCH_BIN(CT, 4,  AC, FD); break; \
                case 8:  LAUNCH_BIN(CT, 8,  AC, FD); break; \
                case 16: LAUNCH_BIN(CT, 16, AC, FD); break; \
                case 32: LAUNCH_BIN(CT, 32, AC, FD); break; \
                case 64: LAUNCH_BIN(CT, 64, AC, FD); break; \
                default: throw std::runtime_error("spmv_binned: lanes 4/8/16/32/64"); }
        #define DISPATCH_BA(CT) \
            if (accum) { if (fuse) { DISPATCH_BL(CT, true, true) } else { DISPATCH_BL(CT, true, false) } } \
            else       { if (fuse) { DISPATCH_BL(CT, false, true) } else { DISPATCH_BL(CT, false, false) } }
        if (col64) { DISPATCH_BA(long) } else { DISPATCH_BA(int) }
        #undef DISPATCH_BA
        #undef DISPATCH_BL
        #undef LAUNCH_BIN
        if (fuse) poff += blocks;
    }
    check_hip("spmv_binned");
    if (fuse && poff > 0)
        reduce_partials(partials, (int)poff, scal, dotslot, dot_accum, stream);
}

void spmv_sell(long nslices, long nrows, long rowbase, uintptr_t sellptr,
               uintptr_t cols, int col64, uintptr_t vals, uintptr_t x,
               uintptr_t y, bool accum, uintptr_t partials, uintptr_t scal,
               int dotslot, bool dot_accum, int variant, uintptr_t perm,
               uintptr_t stream) {
    if (nrows == 0) return;
    long blocks = (nslices * WAVE + BLOCK - 1) / BLOCK;
    if (blocks > MAXG) blocks = MAXG;
    const bool fuse = partials != 0 && dotslot >= 0;
    dim3 g((unsigned)blocks), b(BLOCK);
    if (perm != 0) {
        // sigma-sorted SELL (irregular rows): fixed NT, no swizzle;
        // UNROLL honours the U8 variant bit (deeper unroll = more
        // outstanding x gathers to hide the random-access latency that
        // dominates this path)
        #define LAUNCH_PERM_U(CT, AC, FD, U) \
            hipLaunchKernelGGL((k_spmv_sell<CT, AC, FD, true, false, U, true>), \
                g, b, 0, S(stream), nslices, nrows, rowbase, \
                (const long*)sellptr, (const CT*)cols, (const double*)vals, \
                (const double*)x, (double*)y, (double*)partials, (const int*)perm)
        #define LAUNCH_PERM(CT, AC, FD) \
            if (variant & SELL_U8) { LAUNCH_PERM_U(CT, AC, FD, 8); } \
            else { LAUNCH_PERM_U(CT, AC, FD, 4); }
        #define DISPP(CT) \
            if (accum) { if (fuse) { LAUNCH_PERM(CT, true, true); } else { LAUNCH_PERM(CT, true, false); } } \
            else       { if (fuse) { LAUNCH_PERM(CT, false, true); } else { LAUNCH_PERM(CT, false, false); } }
        if (col64) { DISPP(long) } else { DISPP(int) }
        #undef DISPP
        #undef LAUNCH_PERM
        #undef LAUNCH_PERM_U
        check_hip("spmv_sell_perm");
        if (fuse)
            reduce_partials(partials, (int)blocks, scal, dotslot, dot_accum, stream);
        return;
    }
    #define LAUNCH_SELL(CT, AC, FD, NT, SWZ, U) \
        hipLaunchKernelGGL((k_spmv_sell<CT, AC, FD, NT, SWZ, U>), g, b, 0, S(stream), \
            nslices, nrows, rowbase, (const long*)sellptr, (const CT*)cols, \
            (const double*)vals, (const double*)x, (double*)y, (double*)partials)
    #define DISP_V(CT, AC, FD) \
        switch (variant & (SELL_NT | SELL_SWZ | SELL_U8)) { \
            case 0:                               LAUNCH_SELL(CT, AC, FD, false, false, 4); break; \
            case SELL_NT:                         LAUNCH_SELL(CT, AC, FD, true,  false, 4); break; \
            case SELL_SWZ:                        LAUNCH_SELL(CT, AC, FD, false, true,  4); break; \
            case SELL_NT | SELL_SWZ:              LAUNCH_SELL(CT, AC, FD, true,  true,  4); break; \
            case SELL_U8:                         LAUNCH_SELL(CT, AC, FD, false, false, 8); break; \
            case SELL_U8 | SELL_NT:               LAUNCH_SELL(CT, AC, FD, true,  false, 8); break; \
            case SELL_U8 | SELL_SWZ:              LAUNCH_SELL(CT, AC, FD, false, true,  8); break; \
            default:                              LAUNCH_SELL(CT, AC, FD, true,  true,  8); break; }
    #define DISP2(CT) \
        if (accum) { if (fuse) { DISP_V(CT, true, true) } else { DISP_V(CT, true, false) } } \
        else       { if (fuse) { DISP_V(CT, false, true) } else { DISP_V(CT, false, false) } }
    if (col64) { DISP2(long) } else { DISP2(int) }
    #undef DISP2
    #undef DISP_V
    #undef LAUNCH_SELL
    check_hip("spmv_sell");
    if (fuse)
        reduce_partials(partials, (int)blocks, scal, dotslot, dot_accum, stream);
}

void zero_scalars(uintptr_t scal, int i0, int count, uintptr_t stream) {
    hipLaunchKernelGGL(k_zero_scalars, dim3(1), dim3(BLOCK), 0, S(stream),
                       (double*)scal, i0, count);
    check_hip("zero_scalars");
}

void cg_prep_pt(uintptr_t scal, uintptr_t stream) {
    hipLaunchKernelGGL(k_cg_prep_pt, dim3(1), dim3(64), 0, S(stream), (double*)scal);
    check_hip("cg_prep_pt");
}

void cg_prep_rr(uintptr_t scal, uintptr_t stream) {
    hipLaunchKernelGGL(k_cg_prep_rr, dim3(1), dim3(64), 0, S(stream), (double*)scal);
    check_hip("cg_prep_rr");
}

void dot(uintptr_t x, uintptr_t y, long n, uintptr_t partials, uintptr_t scal,
         int slot, bool accumulate, uintptr_t stream) {
    long blocks = elem_grid(n);
    hipLaunchKernelGGL(k_dot, dim3((unsigned)blocks), dim3(BLOCK), 0, S(stream),
                       (const double*)x, (const double*)y, n, (double*)partials);
    check_hip("dot");
    reduce_partials(partials, (int)blocks, scal, slot, accumulate, stream);
}

void dot2(uintptr_t r, uintptr_t w, long n, uintptr_t partials, uintptr_t scal,
          bool accumulate, uintptr_t stream) {
    long blocks = elem_grid(n);
    hipLaunchKernelGGL(k_dot2, dim3((unsigned)blocks), dim3(BLOCK), 0, S(stream),
                       (const double*)r, (const double*)w, n, (double*)partials);
    check_hip("dot2");
    hipLaunchKernelGGL(k_reduce_partials2, dim3(1), dim3(BLOCK), 0, S(stream),
                       (const double*)partials, (int)blocks, (double*)scal,
                       accumulate ? 1 : 0);
    check_hip("dot2_reduce");
}

void axpy_ratio(uintptr_t y, uintptr_t x, long n, uintptr_t scal, int num, int den,
                double sign, uintptr_t stream) {
    hipLaunchKernelGGL(k_axpy_ratio, dim3((unsigned)elem_grid(n)), dim3(BLOCK), 0, S(stream),
                       (double*)y, (const double*)x, n, (const double*)scal, num, den, sign);
    check_hip("axpy_ratio");
}

void daypx_ratio(uintptr_t y, uintptr_t x, long n, uintptr_t scal, int num, int den,
                 uintptr_t stream) {
    hipLaunchKernelGGL(k_daypx_ratio, dim3((unsigned)elem_grid(n)), dim3(BLOCK), 0, S(stream),
                       (double*)y, (const double*)x, n, (const double*)scal, num, den);
    check_hip("daypx_ratio");
}

void pcg_fused_update(uintptr_t r, uintptr_t x, uintptr_t p, uintptr_t t,
                      uintptr_t z, uintptr_t dinv, long n, uintptr_t scal,
                      uintptr_t partials, uintptr_t stream) {
    long blocks = elem_grid(n);
    hipLaunchKernelGGL(k_pcg_fused_update, dim3((unsigned)blocks), dim3(BLOCK),
                       0, S(stream), (double*)r, (double*)x, (const double*)p,
                       (const double*)t, (double*)z, (const double*)dinv,
                       n, (const double*)scal, (double*)partials);
    check_hip("pcg_fused_update");
    hipLaunchKernelGGL(k_pcg_finalize, dim3(1), dim3(BLOCK), 0, S(stream),
                       (const double*)partials, (int)blocks, (double*)scal);
    check_hip("pcg_finalize");
}

void cg_fused_update(uintptr_t r, uintptr_t x, uintptr_t p, uintptr_t t, long n,
                     uintptr_t scal, uintptr_t partials, uintptr_t stream) {
    long blocks = elem_grid(n);
    hipLaunchKernelGGL(k_cg_fused_update, dim3((unsigned)blocks), dim3(BLOCK), 0, S(stream),
                       (double*)r, (double*)x, (const double*)p, (const double*)t,
                       n, (const double*)scal, (double*)partials);
    check_hip("cg_fused_update");
    hipLaunchKernelGGL(k_cg_finalize, dim3(1), dim3(BLOCK), 0, S(stream),
                       (const double*)partials, (int)blocks, (double*)scal);
    check_hip("cg_finalize");
}

long sell_pipe(long nslices, long nrows_pass, long rowbase, long border_base,
               uintptr_t sellptr, uintptr_t cols, uintptr_t vals,
               uintptr_t w_old, uintptr_t qpart, uintptr_t z, uintptr_t t,
               uintptr_t p, uintptr_t x, uintptr_t r, uintptr_t w_new,
               uintptr_t scal, int first, uintptr_t partials,
               long partials_off, bool mato, uintptr_t stream) {
    if (nrows_pass == 0) return 0;
    long blocks = (nslices * WAVE + BLOCK - 1) / BLOCK;
    // the matA pass (partials_off == 0) must leave partial slots for the
    // matO pass: at >=3.9M rows an uncapped matA claims all MAXG blocks
    // and matO would launch with grid 0 (invalid configuration)
    const long cap = (partials_off == 0) ? (MAXG - 1024) : (MAXG - partials_off);
    if (blocks > cap) blocks = cap;
    dim3 g((unsigned)blocks), b(BLOCK);
    #define LP(MATO) \
        hipLaunchKernelGGL((k_sell_pipe<int, true, 8, MATO>), g, b, 0, S(stream), \
            nslices, nrows_pass, rowbase, border_base, (const long*)sellptr, \
            (const int*)cols, (const double*)vals, (const double*)w_old, \
            (double*)qpart, (double*)z, (double*)t, (double*)p, (double*)x, \
            (double*)r, (double*)w_new, (const double*)scal, first, \
            (double*)partials, partials_off)
    if (mato) { LP(true); } else { LP(false); }
    #undef LP
    check_hip("sell_pipe");
    return blocks;
}

void pipelined_finalize(uintptr_t partials, int nblocks, uintptr_t scal, int first,
                        uintptr_t stream) {
    hipLaunchKernelGGL(k_pipelined_finalize, dim3(1), dim3(BLOCK), 0, S(stream),
                       (const double*)partials, nblocks, (double*)scal, first);
    check_hip("pipelined_finalize");
}

void pipelined_fused(uintptr_t z, uintptr_t t, uintptr_t p, uintptr_t x, uintptr_t r,
                     uintptr_t w, uintptr_t q, long n, uintptr_t scal, int first,
                     uintptr_t partials, bool nt_update, uintptr_t stream) {
    long blocks = elem_grid(n);
    if (nt_update)
        hipLaunchKernelGGL(k_pipelined_fused<true>, dim3((unsigned)blocks), dim3(BLOCK), 0, S(stream),
                           (double*)z, (double*)t, (double*)p, (double*)x, (double*)r,
                           (double*)w, (const double*)q, n, (const double*)scal, first,
                           (double*)partials);
    else
        hipLaunchKernelGGL(k_pipelined_fused<false>, dim3((unsigned)blocks), dim3(BLOCK), 0, S(stream),
                           (double*)z, (double*)t, (double*)p, (double*)x, (double*)r,
                           (double*)w, (const double*)q, n, (const double*)scal, first,
                           (double*)partials);
    check_hip("pipelined_fused");
    hipLaunchKernelGGL(k_pipelined_finalize, dim3(1), dim3(BLOCK), 0, S(stream),
                       (const double*)partials, (int)blocks, (double*)scal, first);
    check_hip("pipelined_finalize");
}

// cooperative launch of the monolithic device CG (reference
// acgsolverhip_solve_device geometry logic, cg-kernels-hip.hip:1892-1909:
// grid = CUs x occupancy, everything grid-stride inside)
int cg_device(long nslices, long nrows, uintptr_t sellptr, uintptr_t cols,
              int col64, uintptr_t vals, uintptr_t b, uintptr_t x, uintptr_t r,
              uintptr_t p, uintptr_t t, uintptr_t scal, uintptr_t partials,
              uintptr_t out2, uintptr_t barrier_state,
              int maxits, double res_atol, double res_rtol,
              uintptr_t stream, int hier) {
    int dev = 0;
    hipGetDevice(&dev);
    hipDeviceProp_t props;
    hipGetDeviceProperties(&props, dev);
    const void* kern = col64 ? (const void*)&k_cg_device<long>
                             : (const void*)&k_cg_device<int>;
    int blocks_per_cu = 0;
    hipError_t oe = hipOccupancyMaxActiveBlocksPerMultiprocessor(
        &blocks_per_cu, kern, BLOCK, 0);
    if (oe != hipSuccess || blocks_per_cu < 1) blocks_per_cu = 1;
    // the occupancy API can over-report by one block/CU and the cooperative
    // launch validates against the same (wrong) number -- a non-resident
    // block would deadlock the grid barrier.  Guide guidance: <= 4 blocks
    // of 256 threads per CU is reliably resident.
    if (blocks_per_cu > 4) blocks_per_cu = 4;
    long grid = (long)props.multiProcessorCount * blocks_per_cu;
    long need = (nslices * WAVE + BLOCK - 1) / BLOCK;
    if (grid > need) grid = need;
    if (grid > MAXG) grid = MAXG;
    // hier < 0 = auto: per-XCD staging pays beyond ~64 blocks (measured
    // MI355X: 1024 blocks flat 192 us/it vs hier 76; 16 blocks flat
    // 15.6 vs 18.7 -- tools/devcg_barrier_ab.py)
    if (hier < 0) hier = grid >= 64 ? 1 : 0;
    // the occupancy API can over-report by one block/CU (guide §1); the
    // cooperative launch checks residency -- shrink and retry on rejection.
    for (;;) {
        void* args[] = {&nslices, &nrows, (void*)&sellptr, (void*)&cols,
                        (void*)&vals, (void*)&b, (void*)&x, (void*)&r,
                        (void*)&p, (void*)&t, (void*)&scal, (void*)&partials,
                        (void*)&out2, (void*)&barrier_state,
                        &maxits, &res_atol, &res_rtol, &hier};
        hipError_t e = hipLaunchCooperativeKernel(
            kern, dim3((unsigned)grid), dim3(BLOCK), args, 0, S(stream));
        if (e == hipSuccess) break;
        if (e == hipErrorCooperativeLaunchTooLarge && grid > props.multiProcessorCount) {
            (void)hipGetLastError();
            grid -= props.multiProcessorCount;
            continue;
        }
        throw std::runtime_error(std::string("cg_device launch: ") + hipGetErrorString(e));
    }
    check_hip("cg_device");
    return (int)grid;
}

void atomic_probe(uintptr_t y, uintptr_t idx, uintptr_t v, long nnz,
                  int mode, uintptr_t stream) {
    const long blocks = elem_grid(nnz, 8);
    hipLaunchKernelGGL(k_atomic_probe, dim3((unsigned)blocks), dim3(BLOCK), 0,
                       S(stream), (double*)y, (const int*)idx,
                       (const double*)v, nnz, mode);
    check_hip("atomic_probe");
}

void pack_gather(uintptr_t sendbuf, uintptr_t x, uintptr_t idx, int idx64, long n,
                 uintptr_t stream) {
    if (n == 0) return;
    dim3 g((unsigned)elem_grid(n, 1)), b(BLOCK);
    if (idx64)
        hipLaunchKernelGGL(k_pack_gather<long>, g, b, 0, S(stream),
                           (double*)sendbuf, (const double*)x, (const long*)idx, n);
    else
        hipLaunchKernelGGL(k_pack_gather<int>, g, b, 0, S(stream),
                           (double*)sendbuf, (const double*)x, (const int*)idx, n);
    check_hip("pack_gather");
}

PYBIND11_MODULE(_acg_kernels, m) {
    m.doc() = "acg_amd gfx950 HIP kernels";
    m.def("spmv", &spmv);
    m.def("spmv_binned", &spmv_binned);
    m.def("spmv_sell", &spmv_sell);
    m.def("zero_scalars", &zero_scalars);
    m.def("cg_prep_pt", &cg_prep_pt);
    m.def("cg_prep_rr", &cg_prep_rr);
    m.def("dot", &dot);
    m.def("dot2", &dot2);
    m.def("axpy_ratio", &axpy_ratio);
    m.def("daypx_ratio", &daypx_ratio);
    m.def("cg_fused_update", &cg_fused_update);
    m.def("pcg_fused_update", &pcg_fused_update);
    m.def("pipelined_fused", &pipelined_fused);
    m.def("sell_pipe", &sell_pipe);
    m.def("pipelined_finalize", &pipelined_finalize);
    m.def("pack_gather", &pack_gather);
    m.def("atomic_probe", &atomic_probe);
    m.def("cg_device", &cg_device);
    m.def("stencil_rowlen", &stencil_rowlen);
    m.def("stencil_fill", &stencil_fill);
    m.def("spmv_bsell", &spmv_bsell);
    m.def("spmv_bsell_daypx", &spmv_bsell_daypx);
    m.def("stencil_blocklen", &stencil_blocklen);
    m.def("stencil_bfill", &stencil_bfill);
    m.def("stencil_spmv", &stencil_spmv);
    m.def("stencil_pipe", &stencil_pipe);
    m.def("stencil_spmv7", &stencil_spmv7);
    m.def("stencil_pipe7", &stencil_pipe7);
    m.attr("S_RR") = S_RR;
    m.attr("S_PT") = S_PT;
    m.attr("S_RR_PREV") = S_RR_PREV;
    m.attr("S_BNRM2") = S_BNRM2;
    m.attr("S_GAMMA") = S_GAMMA;
    m.attr("S_DELTA") = S_DELTA;
    m.attr("S_GAMMA_PREV") = S_GAMMA_PREV;
    m.attr("S_ALPHA_PREV") = S_ALPHA_PREV;
    m.attr("S_NSLOTS") = S_NSLOTS;
    m.attr("MAXG") = MAXG;
}
