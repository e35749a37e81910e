// acg_amd native host preprocessing (C++/OpenMP).
//
// Reference analogs: acg/sort.{c,h} (OpenMP LSD radix sorts, sort.h:82-304),
// acg/prefixsum.{c,h}, and the CSR assembly inside acg/symcsrmatrix.c
// (_init_real_double COO assembly, symcsrmatrix.c:66; packed->full
// conversion _dsymv_init, symcsrmatrix.c:760-845).
//
// These replace the numpy lexsort-based paths for large matrices: the
// symmetric-expansion of a 330M-entry packed operator is O(nnz) with
// parallel counting + atomic scatter + per-row sort instead of an
// O(nnz log nnz) single-threaded lexsort.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <algorithm>
#include <atomic>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

namespace py = pybind11;

using i64 = std::int64_t;

// ---------------------------------------------------------------------------
// prefix sum (reference acgprefixsum_inplace_int64_t, prefixsum.h:94)
static void exclusive_scan(i64* a, i64 n) {
    i64 run = 0;
    for (i64 i = 0; i < n; ++i) {
        i64 v = a[i];
        a[i] = run;
        run += v;
    }
}

// ---------------------------------------------------------------------------
// LSD radix sort returning the sorting permutation
// (reference acgradixsort_int64_t / acgradixsortpair_*, sort.h:82-304)
static void radixsort_perm_impl(const i64* keys, i64 n, i64* perm) {
    std::vector<i64> tmpperm(n);
    std::vector<i64> cnt(256);
    for (i64 i = 0; i < n; ++i) perm[i] = i;
    i64* src = perm;
    i64* dst = tmpperm.data();
    for (int shift = 0; shift < 64; shift += 8) {
        std::fill(cnt.begin(), cnt.end(), 0);
        bool any = false;
        for (i64 i = 0; i < n; ++i) {
            unsigned b = (unsigned)((keys[src[i]] >> shift) & 0xff);
            cnt[b]++;
            any |= b != 0;
        }
        if (!any && shift > 0) continue;
        exclusive_scan(cnt.data(), 256);
        for (i64 i = 0; i < n; ++i) {
            unsigned b = (unsigned)((keys[src[i]] >> shift) & 0xff);
            dst[cnt[b]++] = src[i];
        }
        std::swap(src, dst);
    }
    if (src != perm) std::memcpy(perm, src, n * sizeof(i64));
}

py::array_t<i64> radixsort_perm(py::array_t<i64, py::array::c_style | py::array::forcecast> keys) {
    i64 n = (i64)keys.shape(0);
    py::array_t<i64> perm(n);
    radixsort_perm_impl(keys.data(), n, perm.mutable_data());
    return perm;
}

// ---------------------------------------------------------------------------
// packed-upper symmetric CSR -> full CSR (both triangles), eps on diagonal.
// Parallel: column counting with atomics, scatter with atomic cursors,
// per-row sort by column.
py::tuple sym_expand_full(i64 n,
                          py::array_t<i64, py::array::c_style | py::array::forcecast> rowptr_u,
                          py::array_t<i64, py::array::c_style | py::array::forcecast> col_u,
                          py::array_t<double, py::array::c_style | py::array::forcecast> val_u,
                          double eps, bool col32) {
    const i64* rp = rowptr_u.data();
    const i64* cu = col_u.data();
    const double* vu = val_u.data();
    const i64 nnz_u = rp[n];

    // validate column range before the counting scatter (flag, not throw:
    // throwing inside an OpenMP region is std::terminate)
    std::atomic<i64> badidx(-1);
    #pragma omp parallel for schedule(static)
    for (i64 k = 0; k < nnz_u; ++k) {
        if (cu[k] < 0 || cu[k] >= n) {
            i64 expect = -1;
            badidx.compare_exchange_strong(expect, k, std::memory_order_relaxed);
        }
    }
    if (badidx.load() >= 0) {
        i64 k = badidx.load();
        throw std::out_of_range(
            "sym_expand_full: column " + std::to_string(cu[k]) +
            " at entry " + std::to_string(k) + " outside [0," +
            std::to_string(n) + ")");
    }
    std::vector<std::atomic<i64>> cnt(n);
    for (i64 i = 0; i < n; ++i) cnt[i].store(0, std::memory_order_relaxed);
    #pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        i64 upper = rp[i + 1] - rp[i];
        cnt[i].fetch_add(upper, std::memory_order_relaxed);
        for (i64 k = rp[i]; k < rp[i + 1]; ++k) {
            i64 j = cu[k];
            if (j != i) cnt[j].fetch_add(1, std::memory_order_relaxed);
        }
    }
    py::array_t<i64> rowptr_f(n + 1);
    i64* rf = rowptr_f.mutable_data();
    rf[0] = 0;
    for (i64 i = 0; i < n; ++i) rf[i + 1] = rf[i] + cnt[i].load(std::memory_order_relaxed);
    const i64 nnz_f = rf[n];

    py::array_t<double> vals_f(nnz_f);
    double* vf = vals_f.mutable_data();
    // column index array: int32 or int64
    py::array cols_f;
    void* cfv;
    if (col32) {
        auto a = py::array_t<std::int32_t>(nnz_f);
        cfv = a.mutable_data();
        cols_f = a;
    } else {
        auto a = py::array_t<i64>(nnz_f);
        cfv = a.mutable_data();
        cols_f = a;
    }
    for (i64 i = 0; i < n; ++i) cnt[i].store(0, std::memory_order_relaxed);

    auto scatter = [&](i64 row, i64 col, double v) {
        i64 pos = rf[row] + cnt[row].fetch_add(1, std::memory_order_relaxed);
        vf[pos] = v;
        if (col32) ((std::int32_t*)cfv)[pos] = (std::int32_t)col;
        else ((i64*)cfv)[pos] = col;
    };
    #pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        for (i64 k = rp[i]; k < rp[i + 1]; ++k) {
            i64 j = cu[k];
            double v = vu[k];
            if (j == i) {
                scatter(i, i, v + eps);
            } else {
                scatter(i, j, v);
                scatter(j, i, v);
            }
        }
    }
    // per-row sort by column
    #pragma omp parallel
    {
        std::vector<std::pair<i64, double>> buf;
        #pragma omp for schedule(dynamic, 1024)
        for (i64 i = 0; i < n; ++i) {
            i64 b = rf[i], e = rf[i + 1];
            i64 len = e - b;
            if (len <= 1) continue;
            buf.resize(len);
            for (i64 k = 0; k < len; ++k) {
                i64 c = col32 ? (i64)((std::int32_t*)cfv)[b + k] : ((i64*)cfv)[b + k];
                buf[k] = {c, vf[b + k]};
            }
            std::sort(buf.begin(), buf.end(),
                      [](const auto& x, const auto& y) { return x.first < y.first; });
            for (i64 k = 0; k < len; ++k) {
                if (col32) ((std::int32_t*)cfv)[b + k] = (std::int32_t)buf[k].first;
                else ((i64*)cfv)[b + k] = buf[k].first;
                vf[b + k] = buf[k].second;
            }
        }
    }
    (void)nnz_u;
    return py::make_tuple(rowptr_f, cols_f, vals_f);
}

// ---------------------------------------------------------------------------
// COO -> packed-upper CSR with canonicalisation (swap to row<=col) and
// duplicate summing (reference acgsymcsrmatrix_init_real_double,
// symcsrmatrix.c:66).
py::tuple coo_to_sym_csr(i64 n,
                         py::array_t<i64, py::array::c_style | py::array::forcecast> rows,
                         py::array_t<i64, py::array::c_style | py::array::forcecast> cols,
                         py::array_t<double, py::array::c_style | py::array::forcecast> vals) {
    const i64 nnz = (i64)rows.shape(0);
    const i64* ri = rows.data();
    const i64* ci = cols.data();
    const double* vi = vals.data();

    // Bounds-validate BOTH indices before any scatter touches memory, and
    // record violations in a flag instead of throwing: a throw inside an
    // OpenMP region is std::terminate, not a Python exception.
    std::atomic<i64> badidx(-1);
    #pragma omp parallel for schedule(static)
    for (i64 k = 0; k < nnz; ++k) {
        if (ri[k] < 0 || ri[k] >= n || ci[k] < 0 || ci[k] >= n) {
            i64 expect = -1;
            badidx.compare_exchange_strong(expect, k, std::memory_order_relaxed);
        }
    }
    if (badidx.load() >= 0) {
        i64 k = badidx.load();
        throw std::out_of_range(
            "coo_to_sym_csr: entry " + std::to_string(k) + " (" +
            std::to_string(ri[k]) + "," + std::to_string(ci[k]) +
            ") outside [0," + std::to_string(n) + ")");
    }
    std::vector<std::atomic<i64>> cnt(n);
    for (i64 i = 0; i < n; ++i) cnt[i].store(0, std::memory_order_relaxed);
    #pragma omp parallel for schedule(static)
    for (i64 k = 0; k < nnz; ++k) {
        i64 r = ri[k] <= ci[k] ? ri[k] : ci[k];
        cnt[r].fetch_add(1, std::memory_order_relaxed);
    }
    std::vector<i64> rp(n + 1);
    rp[0] = 0;
    for (i64 i = 0; i < n; ++i) rp[i + 1] = rp[i] + cnt[i].load(std::memory_order_relaxed);
    std::vector<i64> tc(nnz);
    std::vector<double> tv(nnz);
    for (i64 i = 0; i < n; ++i) cnt[i].store(0, std::memory_order_relaxed);
    #pragma omp parallel for schedule(static)
    for (i64 k = 0; k < nnz; ++k) {
        i64 r = ri[k], c = ci[k];
        if (r > c) std::swap(r, c);
        i64 pos = rp[r] + cnt[r].fetch_add(1, std::memory_order_relaxed);
        tc[pos] = c;
        tv[pos] = vi[k];
    }
    // per-row sort + dedup (sum duplicates)
    std::vector<i64> outcnt(n);
    #pragma omp parallel
    {
        std::vector<std::pair<i64, double>> buf;
        #pragma omp for schedule(dynamic, 1024)
        for (i64 i = 0; i < n; ++i) {
            i64 b = rp[i], e = rp[i + 1];
            i64 len = e - b;
            buf.resize(len);
            for (i64 k = 0; k < len; ++k) buf[k] = {tc[b + k], tv[b + k]};
            // value tiebreak: the atomic-cursor scatter lands duplicates in
            // racy order -- sorting them canonically makes the duplicate
            // SUM deterministic across runs (fp addition is order-
            // sensitive; a last-ulp assembly difference visibly forks CG
            // trajectories on ill-conditioned systems)
            std::sort(buf.begin(), buf.end(),
                      [](const auto& x, const auto& y) {
                          return x.first != y.first ? x.first < y.first
                                                    : x.second < y.second;
                      });
            i64 m = 0;
            for (i64 k = 0; k < len; ++k) {
                if (m > 0 && buf[m - 1].first == buf[k].first) {
                    buf[m - 1].second += buf[k].second;
                } else {
                    buf[m++] = buf[k];
                }
            }
            for (i64 k = 0; k < m; ++k) {
                tc[b + k] = buf[k].first;
                tv[b + k] = buf[k].second;
            }
            outcnt[i] = m;
        }
    }
    // compact
    py::array_t<i64> rowptr(n + 1);
    i64* rpo = rowptr.mutable_data();
    rpo[0] = 0;
    for (i64 i = 0; i < n; ++i) rpo[i + 1] = rpo[i] + outcnt[i];
    i64 nnz_out = rpo[n];
    py::array_t<i64> colidx(nnz_out);
    py::array_t<double> v(nnz_out);
    i64* co = colidx.mutable_data();
    double* vo = v.mutable_data();
    #pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        std::memcpy(co + rpo[i], tc.data() + rp[i], outcnt[i] * sizeof(i64));
        std::memcpy(vo + rpo[i], tv.data() + rp[i], outcnt[i] * sizeof(double));
    }
    return py::make_tuple(rowptr, colidx, v);
}

// ---------------------------------------------------------------------------
// Block-structure analysis for Block-SELL conversion: a node = dof
// consecutive rows; its block columns are the union of the rows' col/dof
// values.  Rows are sorted by column, so this is a dof-way merge --
// O(nnz), OpenMP-parallel over nodes (the numpy fallback is an
// O(nnz log nnz) single-threaded unique).
// Returns (blocks_per_node i64[nnodes], bcols i64[nblocks] node-major
// ascending, entry_block i64[nnz] = global block index of each entry).
py::tuple bsell_blocks(py::array_t<i64, py::array::c_style | py::array::forcecast> rowptr,
                       py::array_t<i64, py::array::c_style | py::array::forcecast> colidx,
                       int dof) {
    const i64* rp = rowptr.data();
    const i64* ci = colidx.data();
    const i64 nrows = (i64)rowptr.shape(0) - 1;
    if (dof < 2 || dof > 8 || nrows % dof)
        throw std::runtime_error("bsell_blocks: bad dof");
    const i64 nnodes = nrows / dof;
    const i64 nnz = rp[nrows];
    py::array_t<i64> counts_a(nnodes);
    i64* counts = counts_a.mutable_data();

    auto merge_node = [&](i64 nd, i64* out_bcols, i64* entry_block, i64 blockbase) -> i64 {
        i64 pos[8], end[8];
        for (int r = 0; r < dof; ++r) {
            pos[r] = rp[nd * dof + r];
            end[r] = rp[nd * dof + r + 1];
        }
        i64 nb = 0;
        for (;;) {
            i64 bc = INT64_MAX;
            for (int r = 0; r < dof; ++r)
                if (pos[r] < end[r]) bc = std::min(bc, ci[pos[r]] / dof);
            if (bc == INT64_MAX) break;
            if (out_bcols) out_bcols[nb] = bc;
            for (int r = 0; r < dof; ++r)
                while (pos[r] < end[r] && ci[pos[r]] / dof == bc) {
                    if (entry_block) entry_block[pos[r]] = blockbase + nb;
                    ++pos[r];
                }
            ++nb;
        }
        return nb;
    };

    #pragma omp parallel for schedule(static)
    for (i64 nd = 0; nd < nnodes; ++nd)
        counts[nd] = merge_node(nd, nullptr, nullptr, 0);
    std::vector<i64> starts(nnodes + 1);
    starts[0] = 0;
    for (i64 nd = 0; nd < nnodes; ++nd) starts[nd + 1] = starts[nd] + counts[nd];
    const i64 nblocks = starts[nnodes];
    py::array_t<i64> bcols_a(nblocks);
    py::array_t<i64> entry_a(nnz);
    i64* bcols = bcols_a.mutable_data();
    i64* entry = entry_a.mutable_data();
    #pragma omp parallel for schedule(static)
    for (i64 nd = 0; nd < nnodes; ++nd)
        merge_node(nd, bcols + starts[nd], entry, starts[nd]);
    return py::make_tuple(counts_a, bcols_a, entry_a);
}

// ---------------------------------------------------------------------------
// Heavy-edge matching for multilevel partitioning (reference: the METIS
// coarsening stage wrapped by acg/metis.c:80-436).  Greedy sequential:
// visit vertices in the caller's (random) order, match each unmatched
// vertex with its unmatched neighbour of maximum edge weight; ties break
// toward the first-seen neighbour.  match[v] = partner (== v when no
// unmatched neighbour exists).
py::array_t<i64> hem_match(py::array_t<i64, py::array::c_style | py::array::forcecast> rowptr,
                           py::array_t<i64, py::array::c_style | py::array::forcecast> colidx,
                           py::array_t<double, py::array::c_style | py::array::forcecast> wts,
                           py::array_t<i64, py::array::c_style | py::array::forcecast> order) {
    const i64* rp = rowptr.data();
    const i64* ci = colidx.data();
    const double* w = wts.data();
    const i64* ord = order.data();
    const i64 n = (i64)rowptr.shape(0) - 1;
    py::array_t<i64> match_a(n);
    i64* match = match_a.mutable_data();
    for (i64 i = 0; i < n; ++i) match[i] = -1;
    for (i64 k = 0; k < n; ++k) {
        i64 v = ord[k];
        if (v < 0 || v >= n) throw std::out_of_range("hem_match: order");
        if (match[v] >= 0) continue;
        i64 best = -1;
        double bw = -1.0;
        for (i64 e = rp[v]; e < rp[v + 1]; ++e) {
            i64 u = ci[e];
            if (u == v || u < 0 || u >= n || match[u] >= 0) continue;
            if (w[e] > bw) { bw = w[e]; best = u; }
        }
        if (best >= 0) { match[v] = best; match[best] = v; }
        else match[v] = v;
    }
    return match_a;
}

// ---------------------------------------------------------------------------
// Graph contraction for multilevel partitioning: given a (symmetric,
// both-triangle) fine CSR adjacency and a fine->coarse vertex map, build
// the coarse CSR with parallel-edge weights summed and self-loops
// dropped.  Same machinery as coo_to_sym_csr (atomic counting scatter +
// per-row sort + dedup + compact), OpenMP-parallel -- the numpy
// np.unique route costs a full 80M-key sort per level.
py::tuple contract_graph(py::array_t<i64, py::array::c_style | py::array::forcecast> rowptr,
                         py::array_t<i64, py::array::c_style | py::array::forcecast> cols,
                         py::array_t<double, py::array::c_style | py::array::forcecast> wts,
                         py::array_t<i64, py::array::c_style | py::array::forcecast> cmap,
                         i64 nc) {
    const i64* rp = rowptr.data();
    const i64* ci = cols.data();
    const double* w = wts.data();
    const i64* cm = cmap.data();
    const i64 n = (i64)rowptr.shape(0) - 1;

    std::vector<std::atomic<i64>> cnt(nc);
    for (i64 i = 0; i < nc; ++i) cnt[i].store(0, std::memory_order_relaxed);
    #pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        const i64 ciu = cm[i];
        i64 local = 0;
        for (i64 k = rp[i]; k < rp[i + 1]; ++k)
            if (cm[ci[k]] != ciu) ++local;
        if (local) cnt[ciu].fetch_add(local, std::memory_order_relaxed);
    }
    std::vector<i64> rowc(nc + 1);
    rowc[0] = 0;
    for (i64 i = 0; i < nc; ++i)
        rowc[i + 1] = rowc[i] + cnt[i].load(std::memory_order_relaxed);
    const i64 nnz_c = rowc[nc];
    std::vector<i64> tc(nnz_c);
    std::vector<double> tw(nnz_c);
    for (i64 i = 0; i < nc; ++i) cnt[i].store(0, std::memory_order_relaxed);
    #pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        const i64 ciu = cm[i];
        for (i64 k = rp[i]; k < rp[i + 1]; ++k) {
            const i64 cj = cm[ci[k]];
            if (cj == ciu) continue;
            const i64 pos = rowc[ciu] + cnt[ciu].fetch_add(1, std::memory_order_relaxed);
            tc[pos] = cj;
            tw[pos] = w[k];
        }
    }
    std::vector<i64> outcnt(nc);
    #pragma omp parallel
    {
        std::vector<std::pair<i64, double>> buf;
        #pragma omp for schedule(dynamic, 1024)
        for (i64 i = 0; i < nc; ++i) {
            const i64 b = rowc[i], e = rowc[i + 1];
            const i64 len = e - b;
            buf.resize(len);
            for (i64 k = 0; k < len; ++k) buf[k] = {tc[b + k], tw[b + k]};
            // value tiebreak: canonical duplicate order => deterministic
            // edge-weight sums regardless of the scatter race (see
            // coo_to_sym_csr)
            std::sort(buf.begin(), buf.end(),
                      [](const auto& x, const auto& y) {
                          return x.first != y.first ? x.first < y.first
                                                    : x.second < y.second;
                      });
            i64 m = 0;
            for (i64 k = 0; k < len; ++k) {
                if (m > 0 && buf[m - 1].first == buf[k].first)
                    buf[m - 1].second += buf[k].second;
                else
                    buf[m++] = buf[k];
            }
            for (i64 k = 0; k < m; ++k) { tc[b + k] = buf[k].first; tw[b + k] = buf[k].second; }
            outcnt[i] = m;
        }
    }
    py::array_t<i64> rowptr_c(nc + 1);
    i64* rpo = rowptr_c.mutable_data();
    rpo[0] = 0;
    for (i64 i = 0; i < nc; ++i) rpo[i + 1] = rpo[i] + outcnt[i];
    const i64 out_nnz = rpo[nc];
    py::array_t<i64> cols_c(out_nnz);
    py::array_t<double> w_c(out_nnz);
    i64* co = cols_c.mutable_data();
    double* wo = w_c.mutable_data();
    #pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nc; ++i) {
        std::memcpy(co + rpo[i], tc.data() + rowc[i], outcnt[i] * sizeof(i64));
        std::memcpy(wo + rpo[i], tw.data() + rowc[i], outcnt[i] * sizeof(double));
    }
    return py::make_tuple(rowptr_c, cols_c, w_c);
}

// in-place scans (reference acgprefixsum_inplace_*, prefixsum.h:72-116)
py::array_t<i64> prefix_sum(py::array_t<i64, py::array::c_style | py::array::forcecast> a,
                            bool inclusive) {
    i64 n = (i64)a.shape(0);
    py::array_t<i64> out(n);
    const i64* src = a.data();
    i64* dst = out.mutable_data();
    i64 run = 0;
    for (i64 i = 0; i < n; ++i) {
        if (inclusive) { run += src[i]; dst[i] = run; }
        else { dst[i] = run; run += src[i]; }
    }
    return out;
}

int num_threads() {
#ifdef _OPENMP
    return omp_get_max_threads();
#else
    return 1;
#endif
}

PYBIND11_MODULE(_acg_host, m) {
    m.doc() = "acg_amd native host preprocessing (C++/OpenMP)";
    m.def("radixsort_perm", &radixsort_perm);
    m.def("sym_expand_full", &sym_expand_full,
          py::arg("n"), py::arg("rowptr_u"), py::arg("col_u"), py::arg("val_u"),
          py::arg("eps") = 0.0, py::arg("col32") = true);
    m.def("coo_to_sym_csr", &coo_to_sym_csr);
    m.def("prefix_sum", &prefix_sum, py::arg("a"), py::arg("inclusive") = true);
    m.def("bsell_blocks", &bsell_blocks);
    m.def("hem_match", &hem_match);
    m.def("contract_graph", &contract_graph);
    m.def("num_threads", &num_threads);
}
