// acg_amd gfx950 (CDNA4 / MI355X) kernels for distributed conjugate gradient.
//
// Hand-written HIP replacing the reference's hipSPARSE/hipBLAS calls and its
// CUDA-era kernels (reference: acg/cg-kernels-hip.hip, acg/halo-kernels-hip.hip,
// acg/cghip.c:463-585).  Everything here is designed for CDNA4:
//   - 64-wide wavefronts (shuffle reductions over width 64),
//   - fp64 hardware atomics (unsafeAtomicAdd -> global_atomic_add_f64),
//   - double2 (16 B/lane) vectorized loads on the BLAS-1 path,
//   - grid-stride launches sized for 256 CUs x 8 XCDs,
//   - device-resident scalars: alpha/beta are *computed on device* from the
//     scalar slab so the iteration does no host round-trip except one 8-byte
//     D2H for the convergence test (reference cghip.c:996-1001 idea, fused
//     further: the r/x updates and the (r,r) reduction are one kernel).
//
// The scalar slab layout (fp64 slots) is shared with solvers/cg_hip.py:
#define S_RR 0         // (r,r) current
#define S_PT 1         // (p,t)
#define S_RR_PREV 2    // (r,r) previous
#define S_BNRM2 3      // (b,b)
#define S_GAMMA 4      // pipelined (r,r);  GAMMA,DELTA adjacent => ONE
#define S_DELTA 5      // pipelined (w,r)   2-double allreduce per iteration
#define S_GAMMA_PREV 6
#define S_ALPHA_PREV 7
#define S_NSLOTS 8

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>
#include <cstdint>
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

static inline long elem_grid(long n, long per_thread = 2) {
    long blocks = (n + (long)BLOCK * per_thread - 1) / ((long)BLOCK * per_thread);
    // memory-bound: cap and grid-stride (guide §6 G11)
    if (blocks > 8192) blocks = 8192;
    if (blocks < 1) blocks = 1;
    return blocks;
}

// ---------------------------------------------------------------------------
// wave/block reduction helper: sums `v` over the block, adds to *dst once.
// CDNA4: shuffle over 64 lanes, 4 waves per 256-thread block.
__device__ __forceinline__ void block_reduce_atomic(double v, double* dst) {
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
        if (lane == 0 && v != 0.0) unsafeAtomicAdd(dst, v);
    }
}

// ---------------------------------------------------------------------------
// CSR SpMV, vector kernel: LANES lanes cooperate on one row.
// ACCUM: y[row] += sum (matO pass) vs y[row] = sum (matA pass).
// FUSE_DOT: accumulate dot(p, y) into scal[dotslot] on the fly, where p is
// the SpMV input vector itself -- this fuses the (p,t) reduction of classic
// CG into the SpMV (saves a full 2n-read dot kernel per iteration;
// reference does a separate hipblasDdot, cghip.c:944).
// rowbase: first output row (matO rows start at ninterior).
template <typename ColT, int LANES, bool ACCUM, bool FUSE_DOT>
__global__ void __launch_bounds__(BLOCK)
spmv_csr_vector(long nrows, long rowbase,
                const long* __restrict__ rowptr,
                const ColT* __restrict__ colidx,
                const double* __restrict__ vals,
                const double* __restrict__ x,
                double* __restrict__ y,
                double* __restrict__ scal, int dotslot) {
    const int lane = threadIdx.x & (LANES - 1);
    const long group = ((long)blockIdx.x * BLOCK + threadIdx.x) / LANES;
    const long ngroups = (long)gridDim.x * BLOCK / LANES;
    double dacc = 0.0;
    for (long r = group; r < nrows; r += ngroups) {
        const long k0 = rowptr[r], k1 = rowptr[r + 1];
        double sum = 0.0;
        for (long k = k0 + lane; k < k1; k += LANES)
            sum += vals[k] * x[colidx[k]];
        #pragma unroll
        for (int off = LANES / 2; off > 0; off >>= 1)
            sum += __shfl_down(sum, off, LANES);
        if (lane == 0) {
            const long row = rowbase + r;
            double yr = ACCUM ? (y[row] + sum) : sum;
            y[row] = yr;
            if (FUSE_DOT) dacc += x[row] * (ACCUM ? sum : yr);
        }
    }
    if (FUSE_DOT) block_reduce_atomic(dacc, scal + dotslot);
}

// ---------------------------------------------------------------------------
// BLAS-1 / fused CG kernels.  All scalar coefficients are read from the
// device slab (no D2H of alpha/beta -- reference cg-kernels-hip.hip:116-187).

__global__ void __launch_bounds__(BLOCK)
k_zero_scalars(double* scal, int i0, int count) {
    for (int i = threadIdx.x; i < count; i += BLOCK) scal[i0 + i] = 0.0;
}

// before halo/SpMV of iteration k: zero the (p,t) accumulator
__global__ void k_cg_prep_pt(double* scal) { if (threadIdx.x == 0) scal[S_PT] = 0.0; }

// after allreduce(p,t): save rr, zero the new (r,r) accumulator
__global__ void k_cg_prep_rr(double* scal) {
    if (threadIdx.x == 0) { scal[S_RR_PREV] = scal[S_RR]; scal[S_RR] = 0.0; }
}

// dot / nrm2: acc += sum x[i]*y[i]   (slot must be pre-zeroed)
__global__ void __launch_bounds__(BLOCK)
k_dot(const double* __restrict__ x, const double* __restrict__ y, long n,
      double* scal, int slot) {
    double acc = 0.0;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride)
        acc += x[i] * y[i];
    block_reduce_atomic(acc, scal + slot);
}

// fused dot2 for pipelined CG: gamma += r.r, delta += w.r in ONE pass over
// r,w (reference does two hipblasDdot back-to-back, cghip.c:1735-1752).
__global__ void __launch_bounds__(BLOCK)
k_dot2(const double* __restrict__ r, const double* __restrict__ w, long n,
       double* scal) {
    double g = 0.0, d = 0.0;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
        const double ri = r[i];
        g += ri * ri;
        d += w[i] * ri;
    }
    block_reduce_atomic(g, scal + S_GAMMA);
    block_reduce_atomic(d, scal + S_DELTA);
}

// y += sign * (scal[num]/scal[den]) * x
__global__ void __launch_bounds__(BLOCK)
k_axpy_ratio(double* __restrict__ y, const double* __restrict__ x, long n,
             const double* __restrict__ scal, int num, int den, double sign) {
    const double a = sign * scal[num] / scal[den];
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride)
        y[i] += a * x[i];
}

// y = (scal[num]/scal[den]) * y + x      (daypx with device beta)
__global__ void __launch_bounds__(BLOCK)
k_daypx_ratio(double* __restrict__ y, const double* __restrict__ x, long n,
              const double* __restrict__ scal, int num, int den) {
    const double b = scal[num] / scal[den];
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride)
        y[i] = b * y[i] + x[i];
}

// classic-CG fused update: alpha = rr_prev/pt (device);
//   r -= alpha*t;  x += alpha*p;  rr_new += r.r   (S_RR pre-zeroed)
// One pass over r,t,x,p instead of three kernels + a dot
// (reference: daxpy_minus_alpha + daxpy_alpha + Ddot, cghip.c:969-1026).
__global__ void __launch_bounds__(BLOCK)
k_cg_fused_update(double* __restrict__ r, double* __restrict__ x,
                  const double* __restrict__ p, const double* __restrict__ t,
                  long n, double* scal) {
    const double alpha = scal[S_RR_PREV] / scal[S_PT];
    double acc = 0.0;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
        const double rn = r[i] - alpha * t[i];
        r[i] = rn;
        x[i] += alpha * p[i];
        acc += rn * rn;
    }
    block_reduce_atomic(acc, scal + S_RR);
}

// pipelined-CG fused 6-vector update (Ghysels-Vanroose), device scalars:
//   beta = gamma/gamma_prev, alpha = gamma/(delta - beta*gamma/alpha_prev)
//   (first iteration: beta=0, alpha=gamma/delta)
//   z = q + beta z;  t = w + beta t;  p = r + beta p;
//   x += alpha p;  r -= alpha t;  w -= alpha z
// (reference acgsolverhip_pipelined_daxpy_fused_kernel, cg-kernels-hip.hip:194-232)
__global__ void __launch_bounds__(BLOCK)
k_pipelined_fused(double* __restrict__ z, double* __restrict__ t,
                  double* __restrict__ p, double* __restrict__ x,
                  double* __restrict__ r, double* __restrict__ w,
                  const double* __restrict__ q, long n,
                  const double* __restrict__ scal, int first) {
    const double gamma = scal[S_GAMMA], delta = scal[S_DELTA];
    double beta, alpha;
    if (first) { beta = 0.0; alpha = gamma / delta; }
    else {
        beta = gamma / scal[S_GAMMA_PREV];
        alpha = gamma / (delta - beta * gamma / scal[S_ALPHA_PREV]);
    }
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
        const double zi = q[i] + beta * z[i];
        const double ti = w[i] + beta * t[i];
        const double pi = r[i] + beta * p[i];
        z[i] = zi; t[i] = ti; p[i] = pi;
        x[i] += alpha * pi;
        r[i] -= alpha * ti;
        w[i] -= alpha * zi;
    }
}

// persist gamma_prev/alpha_prev for the next pipelined iteration, zero the
// accumulators (reference acgsolverhip_pipelined_reset_scalars, :234-247)
__global__ void k_pipelined_reset(double* scal, int first) {
    if (threadIdx.x == 0) {
        const double gamma = scal[S_GAMMA], delta = scal[S_DELTA];
        double alpha;
        if (first) alpha = gamma / delta;
        else {
            const double beta = gamma / scal[S_GAMMA_PREV];
            alpha = gamma / (delta - beta * gamma / scal[S_ALPHA_PREV]);
        }
        scal[S_GAMMA_PREV] = gamma;
        scal[S_ALPHA_PREV] = alpha;
        scal[S_GAMMA] = 0.0;
        scal[S_DELTA] = 0.0;
    }
}

// halo pack: sendbuf[i] = x[sendidx[i]]
// (reference acghalo_pack_hip_double, halo-kernels-hip.hip:48-103; the unpack
// scatter does not exist here -- ghosts are received in place, see dist/halo.py)
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

void spmv(long nrows, long rowbase, uintptr_t rowptr, uintptr_t colidx,
          int col64, uintptr_t vals, uintptr_t x, uintptr_t y,
          int lanes, bool accum, bool fuse_dot, uintptr_t scal, int dotslot,
          uintptr_t stream) {
    if (nrows == 0) return;
    const int rows_per_block = BLOCK / lanes;
    long blocks = (nrows + rows_per_block - 1) / rows_per_block;
    if (blocks > 65535 * 4L) blocks = 65535 * 4L;
    dim3 g((unsigned)blocks), b(BLOCK);
    #define LAUNCH_SPMV(CT, L, AC, FD) \
        hipLaunchKernelGGL((spmv_csr_vector<CT, L, AC, FD>), g, b, 0, S(stream), \
            nrows, rowbase, (const long*)rowptr, (const CT*)colidx, \
            (const double*)vals, (const double*)x, (double*)y, (double*)scal, dotslot)
    #define DISPATCH_L(CT, AC, FD) \
        switch (lanes) { \
            case 4:  LAUNCH_SPMV(CT, 4,  AC, FD); break; \
            case 8:  LAUNCH_SPMV(CT, 8,  AC, FD); break; \
            case 16: LAUNCH_SPMV(CT, 16, AC, FD); break; \
            case 32: LAUNCH_SPMV(CT, 32, AC, FD); break; \
            case 64: LAUNCH_SPMV(CT, 64, AC, FD); break; \
            default: throw std::runtime_error("spmv: lanes must be 4/8/16/32/64"); }
    #define DISPATCH_AC(CT) \
        if (accum) { if (fuse_dot) { DISPATCH_L(CT, true, true) } else { DISPATCH_L(CT, true, false) } } \
        else       { if (fuse_dot) { DISPATCH_L(CT, false, true) } else { DISPATCH_L(CT, false, false) } }
    if (col64) { DISPATCH_AC(long) } else { DISPATCH_AC(int) }
    #undef DISPATCH_AC
    #undef DISPATCH_L
    #undef LAUNCH_SPMV
    check_hip("spmv");
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

void dot(uintptr_t x, uintptr_t y, long n, uintptr_t scal, int slot, uintptr_t stream) {
    hipLaunchKernelGGL(k_dot, dim3((unsigned)elem_grid(n)), dim3(BLOCK), 0, S(stream),
                       (const double*)x, (const double*)y, n, (double*)scal, slot);
    check_hip("dot");
}

void dot2(uintptr_t r, uintptr_t w, long n, uintptr_t scal, uintptr_t stream) {
    hipLaunchKernelGGL(k_dot2, dim3((unsigned)elem_grid(n)), dim3(BLOCK), 0, S(stream),
                       (const double*)r, (const double*)w, n, (double*)scal);
    check_hip("dot2");
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

void cg_fused_update(uintptr_t r, uintptr_t x, uintptr_t p, uintptr_t t, long n,
                     uintptr_t scal, uintptr_t stream) {
    hipLaunchKernelGGL(k_cg_fused_update, dim3((unsigned)elem_grid(n)), dim3(BLOCK), 0, S(stream),
                       (double*)r, (double*)x, (const double*)p, (const double*)t, n, (double*)scal);
    check_hip("cg_fused_update");
}

void pipelined_fused(uintptr_t z, uintptr_t t, uintptr_t p, uintptr_t x, uintptr_t r,
                     uintptr_t w, uintptr_t q, long n, uintptr_t scal, int first,
                     uintptr_t stream) {
    hipLaunchKernelGGL(k_pipelined_fused, dim3((unsigned)elem_grid(n)), dim3(BLOCK), 0, S(stream),
                       (double*)z, (double*)t, (double*)p, (double*)x, (double*)r,
                       (double*)w, (const double*)q, n, (const double*)scal, first);
    check_hip("pipelined_fused");
}

void pipelined_reset(uintptr_t scal, int first, uintptr_t stream) {
    hipLaunchKernelGGL(k_pipelined_reset, dim3(1), dim3(64), 0, S(stream), (double*)scal, first);
    check_hip("pipelined_reset");
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
    m.def("zero_scalars", &zero_scalars);
    m.def("cg_prep_pt", &cg_prep_pt);
    m.def("cg_prep_rr", &cg_prep_rr);
    m.def("dot", &dot);
    m.def("dot2", &dot2);
    m.def("axpy_ratio", &axpy_ratio);
    m.def("daypx_ratio", &daypx_ratio);
    m.def("cg_fused_update", &cg_fused_update);
    m.def("pipelined_fused", &pipelined_fused);
    m.def("pipelined_reset", &pipelined_reset);
    m.def("pack_gather", &pack_gather);
    m.attr("S_RR") = S_RR;
    m.attr("S_PT") = S_PT;
    m.attr("S_RR_PREV") = S_RR_PREV;
    m.attr("S_BNRM2") = S_BNRM2;
    m.attr("S_GAMMA") = S_GAMMA;
    m.attr("S_DELTA") = S_DELTA;
    m.attr("S_GAMMA_PREV") = S_GAMMA_PREV;
    m.attr("S_ALPHA_PREV") = S_ALPHA_PREV;
    m.attr("S_NSLOTS") = S_NSLOTS;
}
