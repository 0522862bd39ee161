// Host-side C++ segment trees for prioritized replay.
//
// Replaces the reference's numba-JIT kernels (opendilab/DI-engine
// ding/utils/segment_tree.py:187,210,247) with a pybind11 extension that is
// always available offline. Semantics: classic binary-heap array layout,
// capacity a power of two, leaves at [capacity, 2*capacity).
//
// Batch APIs operate on numpy arrays to amortize the Python boundary:
//   * setitem_batch(tree, idx, val)  -- update leaves + propagate
//   * reduce(tree, start, end)       -- reduce value over [start, end)
//   * find_prefixsum_idx_batch(tree, prefixsum) -- vectorized sampling
#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>
#include <cstdint>
#include <limits>

namespace py = pybind11;

using Arr = py::array_t<double, py::array::c_style | py::array::forcecast>;
using IArr = py::array_t<int64_t, py::array::c_style | py::array::forcecast>;

enum class Op : int { SUM = 0, MIN = 1, MAX = 2 };

static inline double combine(Op op, double a, double b) {
    switch (op) {
        case Op::SUM: return a + b;
        case Op::MIN: return a < b ? a : b;
        default: return a > b ? a : b;
    }
}

// Set leaves idx[i] (0-based positions within [0, capacity)) to val[i], then
// propagate internal nodes bottom-up.
static void setitem_batch(Arr tree, IArr idx, Arr val, int op_i) {
    Op op = static_cast<Op>(op_i);
    auto t = tree.mutable_unchecked<1>();
    auto ix = idx.unchecked<1>();
    auto v = val.unchecked<1>();
    const int64_t cap = t.shape(0) / 2;
    for (py::ssize_t i = 0; i < ix.shape(0); ++i) {
        int64_t node = ix(i) + cap;
        t(node) = v(i);
        node >>= 1;
        while (node >= 1) {
            t(node) = combine(op, t(2 * node), t(2 * node + 1));
            node >>= 1;
        }
    }
}

static double reduce_range(Arr tree, int64_t start, int64_t end, int op_i, double neutral) {
    Op op = static_cast<Op>(op_i);
    auto t = tree.unchecked<1>();
    const int64_t cap = t.shape(0) / 2;
    double result = neutral;
    int64_t l = start + cap, r = end + cap;  // [l, r)
    while (l < r) {
        if (l & 1) result = combine(op, result, t(l++));
        if (r & 1) result = combine(op, result, t(--r));
        l >>= 1;
        r >>= 1;
    }
    return result;
}

// For a SUM tree: walk down from the root following prefix sums; returns leaf
// positions. Vectorized over a batch of prefix sums (the PER sample hot loop).
static IArr find_prefixsum_idx_batch(Arr tree, Arr prefixsum) {
    auto t = tree.unchecked<1>();
    auto p = prefixsum.unchecked<1>();
    const int64_t cap = t.shape(0) / 2;
    IArr out(p.shape(0));
    auto o = out.mutable_unchecked<1>();
    for (py::ssize_t i = 0; i < p.shape(0); ++i) {
        double remain = p(i);
        int64_t node = 1;
        while (node < cap) {
            int64_t left = 2 * node;
            if (t(left) > remain) {
                node = left;
            } else {
                remain -= t(left);
                node = left + 1;
            }
        }
        o(i) = node - cap;
    }
    return out;
}

PYBIND11_MODULE(_ctree, m) {
    m.doc() = "C++ segment-tree kernels for prioritized replay";
    m.def("setitem_batch", &setitem_batch, "batch leaf update + propagate");
    m.def("reduce_range", &reduce_range, "reduce over [start, end)");
    m.def("find_prefixsum_idx_batch", &find_prefixsum_idx_batch, "batched prefix-sum descent");
}
