// C++17 host mirror of the reference's Rust crate API (see
// include/tnc_host.hpp for the mapping). Compiled into libtnc_hip.so.

#include "../../include/tnc_host.hpp"

#include <algorithm>
#include <cassert>
#include <cmath>
#include <cstring>
#include <queue>
#include <set>
#include <stdexcept>
#include <unordered_map>
#include <unordered_set>

#include "../../include/tnc_hip.h"

namespace tnc {

// ---------------------------------------------------------------------------
// gates (tnc/src/gates.rs:150-556, exact constants)
// ---------------------------------------------------------------------------

static const double kS2 = 1.0 / std::sqrt(2.0);

static std::vector<c128> g2(std::initializer_list<c128> v) {
  return std::vector<c128>(v);
}

std::vector<c128> load_gate(const std::string& name,
                            const std::vector<double>& a) {
  const c128 I(0, 1);
  auto need = [&](std::size_t n) {
    if (a.size() != n)
      throw std::runtime_error("Expected " + std::to_string(n) +
                               " angles for gate '" + name + "'");
  };
  if (name == "x") { need(0); return g2({0, 1, 1, 0}); }
  if (name == "y") { need(0); return g2({0, -I, I, 0}); }
  if (name == "z") { need(0); return g2({1, 0, 0, -1}); }
  if (name == "h") { need(0); return g2({kS2, kS2, kS2, -kS2}); }
  if (name == "t") { need(0); return g2({1, 0, 0, c128(kS2, kS2)}); }
  if (name == "u") {
    need(3);
    double th = a[0], phi = a[1], lam = a[2];
    double s = std::sin(th / 2), c = std::cos(th / 2);
    return g2({c128(c, 0), -std::exp(I * lam) * s, std::exp(I * phi) * s,
               std::exp(I * (phi + lam)) * c});
  }
  if (name == "sx") {
    need(0);
    return g2({c128(.5, .5), c128(.5, -.5), c128(.5, -.5), c128(.5, .5)});
  }
  if (name == "sy") {  // exactly as gates.rs:318-323
    need(0);
    return g2({c128(.5, .5), c128(-.5, -.5), c128(.5, .5), c128(.5, .5)});
  }
  if (name == "sz") { need(0); return g2({1, 0, 0, I}); }
  if (name == "rx") {
    need(1);
    double s = std::sin(a[0] / 2), c = std::cos(a[0] / 2);
    return g2({c, -I * s, -I * s, c});
  }
  if (name == "ry") {
    need(1);
    double s = std::sin(a[0] / 2), c = std::cos(a[0] / 2);
    return g2({c, -s, s, c});
  }
  if (name == "rz") {
    need(1);
    return g2({std::exp(-I * (a[0] / 2.0)), 0, 0, std::exp(I * (a[0] / 2.0))});
  }
  if (name == "cx") {
    need(0);
    return g2({1,0,0,0, 0,1,0,0, 0,0,0,1, 0,0,1,0});
  }
  if (name == "cz") {
    need(0);
    return g2({1,0,0,0, 0,1,0,0, 0,0,1,0, 0,0,0,-1});
  }
  if (name == "swap") {
    need(0);
    return g2({1,0,0,0, 0,0,1,0, 0,1,0,0, 0,0,0,1});
  }
  if (name == "cp") {
    need(1);
    return g2({1,0,0,0, 0,1,0,0, 0,0,1,0, 0,0,0, std::exp(I * a[0])});
  }
  if (name == "iswap") {
    need(0);
    return g2({1,0,0,0, 0,0,I,0, 0,I,0,0, 0,0,0,1});
  }
  if (name == "fsim") {
    need(2);
    c128 aa(std::cos(a[0]), 0), bb(0, -std::sin(a[0]));
    c128 cc = std::exp(c128(0, -a[1]));
    return g2({1,0,0,0, 0,aa,bb,0, 0,bb,aa,0, 0,0,0,cc});
  }
  throw std::runtime_error("Gate '" + name + "' not found.");
}

std::vector<c128> matrix_adjoint(const std::vector<c128>& data,
                                 const std::vector<std::uint64_t>& dims) {
  // permute axes (half..n, 0..half) then conjugate (gates.rs:83-101)
  std::size_t nd = dims.size();
  if (nd == 0) return {std::conj(data[0])};
  std::size_t half = nd / 2;
  std::vector<std::uint64_t> out_dims;
  for (std::size_t i = half; i < nd; ++i) out_dims.push_back(dims[i]);
  for (std::size_t i = 0; i < half; ++i) out_dims.push_back(dims[i]);
  std::vector<std::uint64_t> in_stride(nd, 1);
  for (int i = (int)nd - 2; i >= 0; --i)
    in_stride[i] = in_stride[i + 1] * dims[i + 1];
  // out axis k corresponds to in axis perm[k]
  std::vector<std::size_t> perm;
  for (std::size_t i = half; i < nd; ++i) perm.push_back(i);
  for (std::size_t i = 0; i < half; ++i) perm.push_back(i);
  std::uint64_t total = 1;
  for (auto d : dims) total *= d;
  std::vector<c128> out(total);
  std::vector<std::uint64_t> out_stride(nd, 1);
  for (int i = (int)nd - 2; i >= 0; --i)
    out_stride[i] = out_stride[i + 1] * out_dims[i + 1];
  for (std::uint64_t p = 0; p < total; ++p) {
    std::uint64_t src = 0, rem = p;
    for (std::size_t k = 0; k < nd; ++k) {
      std::uint64_t coord = rem / out_stride[k];
      rem %= out_stride[k];
      src += coord * in_stride[perm[k]];
    }
    out[p] = std::conj(data[src]);
  }
  return out;
}

std::vector<c128> load_gate_adjoint(const std::string& name,
                                    const std::vector<double>& angles) {
  auto m = load_gate(name, angles);
  std::vector<std::uint64_t> dims(m.size() == 4 ? 2 : 4, 2);
  return matrix_adjoint(m, dims);
}

// ---------------------------------------------------------------------------
// TensorData
// ---------------------------------------------------------------------------

TensorData TensorData::from_gate(std::string name, std::vector<double> a,
                                 bool adjoint) {
  TensorData d;
  d.kind = Gate;
  d.gate = std::move(name);
  d.angles = std::move(a);
  d.adjoint_flag = adjoint;
  return d;
}

TensorData TensorData::new_from_data(std::vector<c128> data) {
  TensorData d;
  d.kind = Matrix;
  d.matrix = std::move(data);
  return d;
}

std::vector<c128> TensorData::into_data() const {
  switch (kind) {
    case Matrix:
      return matrix;
    case Gate:
      return adjoint_flag ? load_gate_adjoint(gate, angles)
                          : load_gate(gate, angles);
    default:
      throw std::runtime_error("Cannot convert uncontracted tensor to data");
  }
}

TensorData TensorData::adjoint() const {
  TensorData d = *this;
  if (kind == Gate) d.adjoint_flag = !adjoint_flag;
  else if (kind == Matrix) {
    std::vector<std::uint64_t> dims(matrix.size() == 4 ? 2 : 4, 2);
    d.matrix = matrix_adjoint(matrix, dims);
  }
  return d;
}

// ---------------------------------------------------------------------------
// LeafTensor / CompositeTensor (tensor.rs)
// ---------------------------------------------------------------------------

LeafTensor::LeafTensor(std::vector<EdgeIndex> legs,
                       std::vector<std::uint64_t> dims, TensorData data)
    : legs_(std::move(legs)), dims_(std::move(dims)), data_(std::move(data)) {
  assert(legs_.size() == dims_.size());
}

LeafTensor LeafTensor::new_from_const(std::vector<EdgeIndex> legs,
                                      std::uint64_t dim) {
  std::vector<std::uint64_t> dims(legs.size(), dim);
  return LeafTensor(std::move(legs), std::move(dims));
}

double LeafTensor::size() const {
  double s = 1;
  for (auto d : dims_) s *= (double)d;
  return s;
}

static bool contains(const std::vector<EdgeIndex>& v, EdgeIndex x) {
  return std::find(v.begin(), v.end(), x) != v.end();
}

LeafTensor LeafTensor::difference(const LeafTensor& o) const {
  std::vector<EdgeIndex> l;
  std::vector<std::uint64_t> d;
  for (std::size_t i = 0; i < legs_.size(); ++i)
    if (!contains(o.legs_, legs_[i])) {
      l.push_back(legs_[i]);
      d.push_back(dims_[i]);
    }
  return LeafTensor(std::move(l), std::move(d));
}

LeafTensor LeafTensor::union_with(const LeafTensor& o) const {
  std::vector<EdgeIndex> l = legs_;
  std::vector<std::uint64_t> d = dims_;
  for (std::size_t i = 0; i < o.legs_.size(); ++i)
    if (!contains(legs_, o.legs_[i])) {
      l.push_back(o.legs_[i]);
      d.push_back(o.dims_[i]);
    }
  return LeafTensor(std::move(l), std::move(d));
}

LeafTensor LeafTensor::intersection(const LeafTensor& o) const {
  std::vector<EdgeIndex> l;
  std::vector<std::uint64_t> d;
  for (std::size_t i = 0; i < legs_.size(); ++i)
    if (contains(o.legs_, legs_[i])) {
      l.push_back(legs_[i]);
      d.push_back(dims_[i]);
    }
  return LeafTensor(std::move(l), std::move(d));
}

LeafTensor LeafTensor::symmetric_difference(const LeafTensor& o) const {
  LeafTensor a = difference(o);
  LeafTensor b = o.difference(*this);
  std::vector<EdgeIndex> l = a.legs_;
  std::vector<std::uint64_t> d = a.dims_;
  l.insert(l.end(), b.legs_.begin(), b.legs_.end());
  d.insert(d.end(), b.dims_.begin(), b.dims_.end());
  return LeafTensor(std::move(l), std::move(d));
}

Tensor::Tensor(CompositeTensor t)
    : composite(std::make_shared<CompositeTensor>(std::move(t))) {}

LeafTensor CompositeTensor::external_tensor() const {
  LeafTensor acc;
  for (const auto& t : tensors_) {
    LeafTensor leaf = t.is_leaf() ? *t.leaf : t.composite->external_tensor();
    acc = acc ^ leaf;
  }
  return acc;
}

// ---------------------------------------------------------------------------
// paths (contractionpath.rs:197-215)
// ---------------------------------------------------------------------------

ContractionPath ssa_replace_ordering(const ContractionPath& path) {
  ContractionPath out;
  for (const auto& kv : path.nested)
    out.nested[kv.first] = ssa_replace_ordering(kv.second);
  std::unordered_map<TensorIndex, TensorIndex> hs;
  TensorIndex n = path.toplevel.size() + 1;
  for (const auto& [t0, t1] : path.toplevel) {
    TensorIndex a = hs.count(t0) ? hs[t0] : t0;
    TensorIndex b = hs.count(t1) ? hs[t1] : t1;
    hs.emplace(n++, a);
    out.toplevel.emplace_back(a, b);
  }
  return out;
}

// ---------------------------------------------------------------------------
// cost model (contraction_cost.rs)
// ---------------------------------------------------------------------------

double contract_cost_tensors(const LeafTensor& a, const LeafTensor& b) {
  double s = (a & b).size();
  return ((s - 1.0) * 2.0 + s * 6.0) * (a ^ b).size();
}

double contract_op_cost_tensors(const LeafTensor& a, const LeafTensor& b) {
  return (a | b).size();
}

double contract_size_tensors(const LeafTensor& a, const LeafTensor& b) {
  return (a ^ b).size() + a.size() + b.size();
}

static LeafTensor external_leaf(const Tensor& t) {
  return t.is_leaf() ? *t.leaf : t.composite->external_tensor();
}

std::pair<double, double> contract_path_cost(const std::vector<Tensor>& inputs,
                                             const ContractionPath& path,
                                             bool only_count_ops) {
  auto cost_fn = only_count_ops ? contract_op_cost_tensors
                                : contract_cost_tensors;
  double op_cost = 0, mem_cost = 0;
  std::vector<LeafTensor> views;
  views.reserve(inputs.size());
  for (const auto& t : inputs) views.push_back(external_leaf(t));
  for (const auto& kv : path.nested) {
    auto sub = contract_path_cost(inputs[kv.first].composite->tensors(),
                                  kv.second, only_count_ops);
    op_cost += sub.first;
    mem_cost = std::max(mem_cost, sub.second);
  }
  for (const auto& [i, j] : path.toplevel) {
    op_cost += cost_fn(views[i], views[j]);
    mem_cost = std::max(mem_cost, contract_size_tensors(views[i], views[j]));
    views[i] = views[i] ^ views[j];
  }
  return {op_cost, mem_cost};
}

// ---------------------------------------------------------------------------
// greedy pathfinder (cotengrust.rs semantics; same algorithm as
// tnc_amd/paths.py::_greedy_ssa — pinned by the reference's expected paths)
// ---------------------------------------------------------------------------

static std::vector<std::pair<TensorIndex, TensorIndex>> greedy_ssa(
    const std::vector<LeafTensor>& leaves) {
  std::size_t n = leaves.size();
  std::vector<std::pair<TensorIndex, TensorIndex>> ssa;
  if (n <= 1) return ssa;
  std::unordered_map<TensorIndex, LeafTensor> views;
  std::unordered_map<TensorIndex, double> sizes;
  std::unordered_map<EdgeIndex, std::set<TensorIndex>> leg_nodes;
  for (std::size_t i = 0; i < n; ++i) {
    views.emplace(i, leaves[i]);
    sizes[i] = leaves[i].size();
    for (auto l : leaves[i].legs()) leg_nodes[l].insert(i);
  }
  struct Cand {
    double score;
    std::uint64_t counter;
    TensorIndex i, j;
    LeafTensor tij;
  };
  auto cmp = [](const Cand& a, const Cand& b) {
    if (a.score != b.score) return a.score > b.score;  // min-heap
    return a.counter > b.counter;                      // FIFO tie-break
  };
  std::priority_queue<Cand, std::vector<Cand>, decltype(cmp)> heap(cmp);
  std::uint64_t counter = 0;
  auto push_candidate = [&](TensorIndex i, TensorIndex j) {
    LeafTensor tij = views.at(i) ^ views.at(j);
    double score = tij.size() - (sizes[i] + sizes[j]);
    heap.push({score, counter++, i, j, std::move(tij)});
  };
  std::set<std::pair<TensorIndex, TensorIndex>> seen;
  for (std::size_t i = 0; i < n; ++i)
    for (auto l : leaves[i].legs())
      for (auto j : leg_nodes[l])
        if (j > i && seen.insert({i, j}).second) push_candidate(i, j);

  std::set<TensorIndex> alive;
  for (std::size_t i = 0; i < n; ++i) alive.insert(i);
  TensorIndex next_id = n;
  while (!heap.empty()) {
    Cand c = heap.top();
    heap.pop();
    if (!alive.count(c.i) || !alive.count(c.j)) continue;
    TensorIndex nn = next_id++;
    ssa.emplace_back(c.i, c.j);
    alive.erase(c.i);
    alive.erase(c.j);
    alive.insert(nn);
    sizes[nn] = c.tij.size();
    std::set<TensorIndex> neighbors;
    for (auto l : c.tij.legs()) {
      auto& s = leg_nodes[l];
      s.erase(c.i);
      s.erase(c.j);
      s.insert(nn);
      for (auto x : s)
        if (x != nn && alive.count(x)) neighbors.insert(x);
    }
    for (auto old : {c.i, c.j}) {
      auto it = views.find(old);
      if (it != views.end())
        for (auto l : it->second.legs()) leg_nodes[l].erase(old);
    }
    views.emplace(nn, std::move(c.tij));
    for (auto nb : neighbors) push_candidate(nn, nb);
  }
  // remaining disconnected terms: pairwise outer products smallest-first;
  // ties pop the LARGER ssa id first (cotengrust.rs:262-291)
  struct Rest {
    double size;
    TensorIndex id;
  };
  auto rcmp = [](const Rest& a, const Rest& b) {
    if (a.size != b.size) return a.size > b.size;
    return a.id < b.id;  // pop larger id first on ties
  };
  std::priority_queue<Rest, std::vector<Rest>, decltype(rcmp)> rest(rcmp);
  for (auto id : alive) rest.push({sizes[id], id});
  while (rest.size() > 1) {
    Rest a = rest.top(); rest.pop();
    Rest b = rest.top(); rest.pop();
    TensorIndex nn = next_id++;
    ssa.emplace_back(a.id, b.id);
    LeafTensor tab = views.at(a.id) ^ views.at(b.id);
    sizes[nn] = tab.size();
    views.emplace(nn, std::move(tab));
    rest.push({sizes[nn], nn});
  }
  return ssa;
}

ContractionPathResult Greedy::find_path(const CompositeTensor& tn) {
  ContractionPathResult res;
  std::vector<LeafTensor> leaves;
  for (std::size_t idx = 0; idx < tn.tensors().size(); ++idx) {
    const Tensor& t = tn.tensors()[idx];
    if (t.is_leaf()) {
      leaves.push_back(*t.leaf);
    } else {
      Greedy sub;
      auto r = sub.find_path(*t.composite);
      res.ssa_path.nested[idx] = r.ssa_path;
      leaves.push_back(t.composite->external_tensor());
    }
  }
  res.ssa_path.toplevel = greedy_ssa(leaves);
  auto replace = ssa_replace_ordering(res.ssa_path);
  auto cost = contract_path_cost(tn.tensors(), replace, true);
  res.flops = cost.first;
  res.size = cost.second;
  return res;
}

// ---------------------------------------------------------------------------
// circuit builder (circuit_builder.rs)
// ---------------------------------------------------------------------------

std::size_t Circuit::allocate_register(std::size_t n) {
  std::size_t base = open_edges_.size();
  for (std::size_t q = 0; q < n; ++q) {
    EdgeIndex e = next_edge_++;
    open_edges_.push_back(e);
    LeafTensor ket0({e}, {2},
                    TensorData::new_from_data({c128(1, 0), c128(0, 0)}));
    tn_.push_tensor(Tensor(std::move(ket0)));
  }
  return base;
}

void Circuit::append_gate(TensorData gate,
                          const std::vector<std::size_t>& qubits) {
  // legs = new edges then old edges (circuit_builder.rs:196-220)
  std::vector<EdgeIndex> edges;
  for (std::size_t k = 0; k < qubits.size(); ++k)
    edges.push_back(next_edge_ + k);
  for (auto q : qubits) edges.push_back(open_edges_.at(q));
  for (std::size_t k = 0; k < qubits.size(); ++k)
    open_edges_[qubits[k]] = next_edge_ + k;
  next_edge_ += qubits.size();
  std::vector<std::uint64_t> dims(edges.size(), 2);
  tn_.push_tensor(
      Tensor(LeafTensor(std::move(edges), std::move(dims), std::move(gate))));
}

std::pair<CompositeTensor, std::vector<EdgeIndex>>
Circuit::into_amplitude_network(const std::string& bitstring) {
  assert(bitstring.size() == open_edges_.size());
  std::vector<EdgeIndex> final_legs;
  for (std::size_t q = 0; q < bitstring.size(); ++q) {
    char c = bitstring[q];
    if (c == '*') {
      final_legs.push_back(open_edges_[q]);
      continue;
    }
    std::vector<c128> bra = c == '0'
                                ? std::vector<c128>{c128(1, 0), c128(0, 0)}
                                : std::vector<c128>{c128(0, 0), c128(1, 0)};
    tn_.push_tensor(Tensor(LeafTensor({open_edges_[q]}, {2},
                                      TensorData::new_from_data(bra))));
  }
  return {tn_, final_legs};
}

std::pair<CompositeTensor, std::vector<EdgeIndex>>
Circuit::into_statevector_network() {
  return into_amplitude_network(std::string(open_edges_.size(), '*'));
}

// ---------------------------------------------------------------------------
// contract_tensor_network via the device executor (contraction.rs:35-68)
// ---------------------------------------------------------------------------

namespace {
struct FlatPlan {
  std::vector<const LeafTensor*> leaves;
  std::vector<std::pair<std::uint64_t, std::uint64_t>> steps;
};

std::size_t flatten(const CompositeTensor& tn, const ContractionPath& path,
                    FlatPlan& plan) {
  std::vector<std::size_t> slot;
  for (std::size_t idx = 0; idx < tn.tensors().size(); ++idx) {
    const Tensor& child = tn.tensors()[idx];
    if (child.is_leaf()) {
      slot.push_back(plan.leaves.size());
      plan.leaves.push_back(child.leaf.get());
    } else {
      auto it = path.nested.find(idx);
      if (it == path.nested.end())
        throw std::runtime_error("composite child without nested path");
      slot.push_back(flatten(*child.composite, it->second, plan));
    }
  }
  if (path.toplevel.empty()) {
    if (slot.size() != 1)
      throw std::runtime_error("path does not contract composite");
    return slot[0];
  }
  std::set<std::size_t> consumed;
  for (const auto& [i, j] : path.toplevel) {
    plan.steps.emplace_back(slot.at(i), slot.at(j));
    consumed.insert(slot.at(j));
  }
  std::size_t remaining = SIZE_MAX;
  int count = 0;
  for (auto g : slot)
    if (!consumed.count(g)) {
      remaining = g;
      ++count;
    }
  if (count != 1) throw std::runtime_error("path leaves multiple tensors");
  return remaining;
}
}  // namespace

LeafTensor contract_tensor_network(const CompositeTensor& tn,
                                   const ContractionPath& replace_path,
                                   int device) {
  FlatPlan plan;
  flatten(tn, replace_path, plan);
  tn_net* net = tn_net_create(device);
  if (!net) throw std::runtime_error(std::string(tn_last_error()));
  try {
    for (const LeafTensor* t : plan.leaves) {
      auto data = t->tensor_data().into_data();
      std::uint64_t elems = 1;
      for (auto d : t->bond_dims()) elems *= d;
      if (data.size() != elems)
        throw std::runtime_error("leaf data size mismatch");
      auto idx = tn_net_add_leaf(net, t->legs().data(), t->bond_dims().data(),
                                 t->legs().size(), data.data());
      if (idx < 0) throw std::runtime_error(std::string(tn_last_error()));
    }
    std::vector<std::uint64_t> pairs;
    for (const auto& [i, j] : plan.steps) {
      pairs.push_back(i);
      pairs.push_back(j);
    }
    int rc = tn_net_contract(net, pairs.data(), plan.steps.size(), nullptr);
    if (rc != TN_OK) throw std::runtime_error(std::string(tn_last_error()));
    std::uint64_t labels[64], dims[64];
    std::size_t nd = 0;
    rc = tn_net_result_meta(net, labels, dims, &nd);
    if (rc != TN_OK) throw std::runtime_error(std::string(tn_last_error()));
    std::uint64_t elems = 1;
    std::vector<EdgeIndex> legs(labels, labels + nd);
    std::vector<std::uint64_t> dvec(dims, dims + nd);
    for (auto d : dvec) elems *= d;
    std::vector<c128> out(elems);
    rc = tn_net_result_data(net, out.data());
    if (rc != TN_OK) throw std::runtime_error(std::string(tn_last_error()));
    tn_net_destroy(net);
    return LeafTensor(std::move(legs), std::move(dvec),
                      TensorData::new_from_data(std::move(out)));
  } catch (...) {
    tn_net_destroy(net);
    throw;
  }
}

}  // namespace tnc
