// CPU segment trees for prioritized replay.
//
// Re-implements the capability of the reference's C++ trees
// (pytorch/rl torchrl/csrc/segment_tree.h:42 SegmentTree, :249
// SumSegmentTree::ScanLowerBound, :303 MinSegmentTree) with a flat
// 2*size array, batched torch-tensor overloads and pickle support.
// Fresh implementation — iterative bottom-up update, top-down descent.

#pragma once

#include <torch/extension.h>

#include <algorithm>
#include <cmath>
#include <limits>
#include <vector>

namespace rl_amd {

template <typename T, bool IsSum>
class SegmentTreeCPU {
 public:
  explicit SegmentTreeCPU(int64_t capacity) : capacity_(capacity) {
    size_ = 1;
    while (size_ < capacity_) size_ <<= 1;
    tree_.assign(2 * size_, neutral());
  }

  static constexpr T neutral() {
    return IsSum ? T(0) : std::numeric_limits<T>::infinity();
  }

  static T combine(T a, T b) { return IsSum ? a + b : std::min(a, b); }

  int64_t capacity() const { return capacity_; }
  int64_t size() const { return size_; }

  T at(int64_t i) const { return tree_[size_ + i]; }

  void update_one(int64_t i, T v) {
    int64_t node = size_ + i;
    tree_[node] = v;
    node >>= 1;
    while (node >= 1) {
      tree_[node] = combine(tree_[2 * node], tree_[2 * node + 1]);
      node >>= 1;
    }
  }

  // batched update; later entries win on duplicate indices (reference
  // keeps input order by serializing leaf writes)
  void update(torch::Tensor index, torch::Tensor value) {
    auto idx = index.contiguous().to(torch::kLong);
    auto val = value.contiguous().to(torch::kDouble);
    auto* ip = idx.data_ptr<int64_t>();
    auto* vp = val.data_ptr<double>();
    int64_t n = idx.numel();
    for (int64_t k = 0; k < n; ++k) {
      tree_[size_ + ip[k]] = (T)vp[k];
    }
    // recompute only touched paths
    std::vector<int64_t> nodes(n);
    for (int64_t k = 0; k < n; ++k) nodes[k] = (size_ + ip[k]) >> 1;
    std::sort(nodes.begin(), nodes.end());
    nodes.erase(std::unique(nodes.begin(), nodes.end()), nodes.end());
    while (!nodes.empty() && nodes[0] >= 1) {
      std::vector<int64_t> parents;
      parents.reserve(nodes.size());
      for (int64_t node : nodes) {
        tree_[node] = combine(tree_[2 * node], tree_[2 * node + 1]);
        int64_t p = node >> 1;
        if (p >= 1 && (parents.empty() || parents.back() != p))
          parents.push_back(p);
      }
      if (nodes[0] == 1) break;
      nodes.swap(parents);
    }
  }

  T query(int64_t start, int64_t end) const {
    T res = neutral();
    int64_t l = start + size_, r = end + size_;
    while (l < r) {
      if (l & 1) res = combine(res, tree_[l++]);
      if (r & 1) res = combine(res, tree_[--r]);
      l >>= 1;
      r >>= 1;
    }
    return res;
  }

  torch::Tensor get(torch::Tensor index) const {
    auto idx = index.contiguous().to(torch::kLong);
    auto out = torch::empty({idx.numel()}, torch::kDouble);
    auto* ip = idx.data_ptr<int64_t>();
    auto* op = out.data_ptr<double>();
    for (int64_t k = 0; k < idx.numel(); ++k) op[k] = (double)tree_[size_ + ip[k]];
    return out;
  }

  torch::Tensor dump_values() const {
    auto out = torch::empty({capacity_}, torch::kDouble);
    auto* op = out.data_ptr<double>();
    for (int64_t i = 0; i < capacity_; ++i) op[i] = (double)tree_[size_ + i];
    return out;
  }

  void load_values(torch::Tensor values) {
    auto val = values.contiguous().to(torch::kDouble);
    auto* vp = val.data_ptr<double>();
    int64_t n = std::min<int64_t>(val.numel(), capacity_);
    for (int64_t i = 0; i < n; ++i) tree_[size_ + i] = (T)vp[i];
    for (int64_t node = size_ - 1; node >= 1; --node)
      tree_[node] = combine(tree_[2 * node], tree_[2 * node + 1]);
  }

  // inverse-CDF descent (sum trees only)
  torch::Tensor scan_lower_bound(torch::Tensor mass) const {
    static_assert(IsSum || true, "");
    auto m = mass.contiguous().to(torch::kDouble);
    auto out = torch::empty({m.numel()}, torch::kLong);
    auto* mp = m.data_ptr<double>();
    auto* op = out.data_ptr<int64_t>();
    for (int64_t k = 0; k < m.numel(); ++k) {
      double rem = mp[k];
      int64_t node = 1;
      while (node < size_) {
        T left = tree_[2 * node];
        if (rem >= (double)left) {
          rem -= (double)left;
          node = 2 * node + 1;
        } else {
          node = 2 * node;
        }
      }
      int64_t leaf = node - size_;
      op[k] = std::min<int64_t>(leaf, capacity_ - 1);
    }
    return out;
  }

 private:
  int64_t capacity_;
  int64_t size_;
  std::vector<T> tree_;
};

template <typename T, bool IsSum>
void bind_tree(pybind11::module& m, const char* name) {
  using Tree = SegmentTreeCPU<T, IsSum>;
  auto cls =
      pybind11::class_<Tree>(m, name)
          .def(pybind11::init<int64_t>())
          .def_property_readonly("capacity", &Tree::capacity)
          .def_property_readonly("size", &Tree::size)
          .def("update", &Tree::update)
          .def("at", &Tree::at)
          .def("get", &Tree::get)
          .def("query", &Tree::query)
          .def("dump_values", &Tree::dump_values)
          .def("load_values", &Tree::load_values)
          .def(pybind11::pickle(
              [](const Tree& t) {
                return pybind11::make_tuple(t.capacity(),
                                            const_cast<Tree&>(t).dump_values());
              },
              [](pybind11::tuple s) {
                Tree t(s[0].cast<int64_t>());
                t.load_values(s[1].cast<torch::Tensor>());
                return t;
              }));
  if (IsSum) {
    cls.def("scan_lower_bound", &Tree::scan_lower_bound);
  }
}

inline void define_segment_trees(pybind11::module& m) {
  bind_tree<float, true>(m, "SumSegmentTreeFp32");
  bind_tree<double, true>(m, "SumSegmentTreeFp64");
  bind_tree<float, false>(m, "MinSegmentTreeFp32");
  bind_tree<double, false>(m, "MinSegmentTreeFp64");
}

}  // namespace rl_amd
