// Assertions over the C++ host mirror (include/tnc_host.hpp) against the
// reference's pinned values: leg algebra (tensor.rs doctests), SSA
// conversion (contractionpath.rs:286-294), cost model
// (contraction_cost.rs:324-356), greedy expected paths
// (cotengrust.rs:229-307), gate adjoint consistency (gates.rs:585-607).
// With argv[1] == "gpu": end-to-end contraction on the device (GHZ 1/sqrt2,
// <0|H^5|0> = 2^-2.5, nested composite walk).
#include <cassert>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <map>

#include "../../include/tnc_host.hpp"
#include "../../include/tnc_hip.h"

using namespace tnc;

static LeafTensor mk(std::vector<EdgeIndex> legs,
                     std::map<EdgeIndex, std::uint64_t> bd) {
  std::vector<std::uint64_t> dims;
  for (auto l : legs) dims.push_back(bd.at(l));
  return LeafTensor(legs, dims);
}

static void test_leg_algebra() {
  std::map<EdgeIndex, std::uint64_t> bd{{1, 2}, {2, 4}, {3, 6}, {4, 3}, {5, 9}};
  auto t1 = mk({1, 2, 3}, bd), t2 = mk({4, 2, 5}, bd);
  assert(((t1 - t2).legs() == std::vector<EdgeIndex>{1, 3}));
  assert(((t1 | t2).legs() == std::vector<EdgeIndex>{1, 2, 3, 4, 5}));
  assert(((t1 & t2).legs() == std::vector<EdgeIndex>{2}));
  assert(((t1 ^ t2).legs() == std::vector<EdgeIndex>{1, 3, 4, 5}));
  assert(((t1 ^ t2).bond_dims() == std::vector<std::uint64_t>{2, 6, 3, 9}));
}

static void test_ssa_replace() {
  ContractionPath p = ContractionPath::simple(
      {{0, 3}, {1, 2}, {6, 4}, {5, 7}, {9, 8}, {11, 10}});
  auto r = ssa_replace_ordering(p);
  std::vector<std::pair<TensorIndex, TensorIndex>> expect{
      {0, 3}, {1, 2}, {6, 4}, {5, 0}, {6, 1}, {6, 5}};
  assert(r.toplevel == expect);
}

static void test_cost_model() {
  std::map<EdgeIndex, std::uint64_t> bd{{0, 5}, {1, 7}, {2, 9}, {3, 11},
                                        {4, 13}};
  auto t1 = mk({0, 1, 2}, bd), t2 = mk({2, 3, 4}, bd);
  assert(contract_cost_tensors(t1, t2) == 350350.0);
  assert(contract_op_cost_tensors(t1, t2) == 45045.0);
  assert(contract_size_tensors(t1, t2) == 6607.0);

  std::map<EdgeIndex, std::uint64_t> bd2{{0, 5}, {1, 2}, {2, 6}, {3, 8},
                                         {4, 1}, {5, 3}, {6, 4}};
  CompositeTensor tn({Tensor(mk({4, 3, 2}, bd2)), Tensor(mk({0, 1, 3, 2}, bd2)),
                      Tensor(mk({4, 5, 6}, bd2))});
  auto c = contract_path_cost(tn.tensors(),
                              ContractionPath::simple({{0, 1}, {0, 2}}), false);
  assert(c.first == 4540.0 && c.second == 538.0);
  auto c2 = contract_path_cost(tn.tensors(),
                               ContractionPath::simple({{0, 1}, {0, 2}}), true);
  assert(c2.first == 600.0 && c2.second == 538.0);
}

static void expect_path(const ContractionPathResult& r,
                        std::vector<std::pair<TensorIndex, TensorIndex>> p,
                        double flops, double size) {
  if (r.ssa_path.toplevel != p || r.flops != flops || r.size != size) {
    std::fprintf(stderr, "greedy mismatch: flops=%g size=%g path:", r.flops,
                 r.size);
    for (auto& [a, b] : r.ssa_path.toplevel)
      std::fprintf(stderr, " (%zu,%zu)", a, b);
    std::fprintf(stderr, "\n");
    assert(false);
  }
}

static void test_greedy_pinned() {
  std::map<EdgeIndex, std::uint64_t> bd{{0, 5}, {1, 2}, {2, 6}, {3, 8},
                                        {4, 1}, {5, 3}, {6, 4}};
  CompositeTensor simple({Tensor(mk({4, 3, 2}, bd)),
                          Tensor(mk({0, 1, 3, 2}, bd)),
                          Tensor(mk({4, 5, 6}, bd))});
  Greedy g;
  expect_path(g.find_path(simple), {{0, 1}, {3, 2}}, 600.0, 538.0);

  CompositeTensor inner({Tensor(mk({4, 3, 2}, bd)), Tensor(mk({4, 3, 2}, bd)),
                         Tensor(mk({0, 1, 5}, bd)), Tensor(mk({1, 6}, bd))});
  expect_path(g.find_path(inner), {{0, 1}, {2, 3}, {4, 5}}, 228.0, 121.0);

  std::map<EdgeIndex, std::uint64_t> bd3{{0, 3}, {1, 2}, {2, 2}};
  CompositeTensor outer({Tensor(mk({0}, bd3)), Tensor(mk({1}, bd3)),
                         Tensor(mk({2}, bd3))});
  expect_path(g.find_path(outer), {{2, 1}, {0, 3}}, 16.0, 19.0);

  std::map<EdgeIndex, std::uint64_t> bd4{{0, 5}, {1, 4}};
  CompositeTensor outer2({Tensor(mk({0}, bd4)), Tensor(mk({0}, bd4)),
                          Tensor(mk({1}, bd4)), Tensor(mk({1}, bd4))});
  expect_path(g.find_path(outer2), {{0, 1}, {2, 3}, {5, 4}}, 10.0, 11.0);

  std::map<EdgeIndex, std::uint64_t> bd5{
      {0, 27}, {1, 18}, {2, 12}, {3, 15}, {4, 5},  {5, 3},
      {6, 18}, {7, 22}, {8, 45}, {9, 65}, {10, 5}, {11, 17}};
  CompositeTensor cx({Tensor(mk({4, 3, 2}, bd5)), Tensor(mk({0, 1, 3, 2}, bd5)),
                      Tensor(mk({4, 5, 6}, bd5)), Tensor(mk({6, 8, 9}, bd5)),
                      Tensor(mk({10, 8, 9}, bd5)), Tensor(mk({5, 1, 0}, bd5))});
  expect_path(g.find_path(cx), {{1, 5}, {3, 4}, {6, 0}, {7, 2}, {9, 8}},
              529815.0, 89478.0);
}

static void test_gates() {
  // adjoint == conj-transpose and unitarity for a parametrized gate
  auto u = load_gate("u", {1.1, -0.4, 2.2});
  auto ua = load_gate_adjoint("u", {1.1, -0.4, 2.2});
  for (int i = 0; i < 2; ++i)
    for (int j = 0; j < 2; ++j) {
      c128 s = 0;
      for (int k = 0; k < 2; ++k) s += u[i * 2 + k] * ua[k * 2 + j];
      assert(std::abs(s - c128(i == j ? 1 : 0, 0)) < 1e-14);
    }
  auto sy = load_gate("sy", {});
  assert(sy[1] == c128(-0.5, -0.5) && sy[2] == c128(0.5, 0.5));
}

static void test_gpu_contraction() {
  // GHZ amplitude <000|GHZ> = 1/sqrt(2) (README config 1)
  Circuit c;
  c.allocate_register(3);
  c.append_gate(TensorData::from_gate("h"), {0});
  c.append_gate(TensorData::from_gate("cx"), {0, 1});
  c.append_gate(TensorData::from_gate("cx"), {1, 2});
  auto [tn, perm] = c.into_amplitude_network("000");
  Greedy g;
  auto path = g.find_path(tn).replace_path();
  auto out = contract_tensor_network(tn, path);
  assert(out.legs().empty());
  assert(std::abs(out.tensor_data().matrix[0] - c128(1 / std::sqrt(2.0), 0)) <
         1e-14);

  // <00000|H^5|00000> = 2^-2.5 (circuit_builder.rs:362-385)
  Circuit c2;
  c2.allocate_register(5);
  for (std::size_t q = 0; q < 5; ++q)
    c2.append_gate(TensorData::from_gate("h"), {q});
  auto [tn2, perm2] = c2.into_amplitude_network("00000");
  auto out2 = contract_tensor_network(tn2, g.find_path(tn2).replace_path());
  assert(std::abs(out2.tensor_data().matrix[0] -
                  c128(std::pow(1 / std::sqrt(2.0), 5), 0)) < 1e-14);

  // nested composite walk (contraction.rs:42-49)
  std::map<EdgeIndex, std::uint64_t> bd{{0, 2}, {1, 3}, {2, 4}, {3, 2}};
  LeafTensor a = mk({0, 1}, bd), b = mk({1, 2}, bd), d = mk({2, 3}, bd);
  std::vector<c128> va(6), vb(12), vd(8);
  for (std::size_t i = 0; i < va.size(); ++i) va[i] = c128(0.5 + i, -1.0);
  for (std::size_t i = 0; i < vb.size(); ++i) vb[i] = c128(0.25 * i, 0.5);
  for (std::size_t i = 0; i < vd.size(); ++i) vd[i] = c128(1.0, 0.125 * i);
  a.set_tensor_data(TensorData::new_from_data(va));
  b.set_tensor_data(TensorData::new_from_data(vb));
  d.set_tensor_data(TensorData::new_from_data(vd));
  CompositeTensor innerc({Tensor(a), Tensor(b)});
  CompositeTensor nested({Tensor(innerc), Tensor(d)});
  ContractionPath p;
  p.nested[0] = ContractionPath::simple({{0, 1}});
  p.toplevel = {{0, 1}};
  auto out3 = contract_tensor_network(nested, p);
  assert((out3.legs() == std::vector<EdgeIndex>{0, 3}));
  // spot value vs hand-computed: out[0][0] = sum_{1,2} a[0][l1] b[l1][l2] d[l2][0]
  c128 expect = 0;
  for (int l1 = 0; l1 < 3; ++l1)
    for (int l2 = 0; l2 < 4; ++l2)
      expect += va[l1] * vb[l1 * 4 + l2] * vd[l2 * 2 + 0];
  assert(std::abs(out3.tensor_data().matrix[0] - expect) < 1e-12);
}

int main(int argc, char** argv) {
  test_leg_algebra();
  test_ssa_replace();
  test_cost_model();
  test_greedy_pinned();
  test_gates();
  std::printf("host mirror CPU assertions OK\n");
  if (argc > 1 && std::strcmp(argv[1], "gpu") == 0) {
    if (tn_device_count() == 0) {
      std::fprintf(stderr, "no GPU\n");
      return 2;
    }
    test_gpu_contraction();
    std::printf("host mirror GPU contraction OK\n");
  }
  return 0;
}
