// tnc_host — C++17 host-side mirror of the reference's Rust crate API.
//
// The north star keeps host orchestration and path/partition search in the
// reference's host language; the reference is compiled Rust (no Rust
// toolchain in this image), so this mirror carries the same names and
// semantics in C++. Python (tnc_amd/) wraps the same
// functionality for tests and the benchmark; this header is the
// compiled-host entry point a Rust/C++ integrator would use.
//
// Mirrored API (reference citations):
//   LeafTensor / CompositeTensor / TensorData      tensor.rs, tensordata.rs
//   leg set algebra (^ & | -)                      tensor.rs:629-725
//   ContractionPath {nested, toplevel}             contractionpath.rs:30-35
//   ssa_replace_ordering                           contractionpath.rs:197-215
//   contract_cost_tensors / op / size              contraction_cost.rs:26-91
//   contract_path_cost                             contraction_cost.rs:101-151
//   Pathfinder (Greedy)                            paths.rs:21-28, cotengrust.rs
//   load_gate / load_gate_adjoint (18 gates)       gates.rs:17-38, 150-556
//   Circuit builder (+amplitude/statevector nets)  circuit_builder.rs:124-327
//   contract_tensor_network (device executor)      contraction.rs:35-68

#ifndef TNC_HOST_HPP
#define TNC_HOST_HPP

#include <complex>
#include <cstdint>
#include <map>
#include <memory>
#include <string>
#include <utility>
#include <vector>

namespace tnc {

using EdgeIndex = std::uint64_t;
using TensorIndex = std::size_t;
using c128 = std::complex<double>;

// --- TensorData (tensordata.rs:17-27; File variant out of scope) ---
struct TensorData {
  enum Kind { None, Gate, Matrix } kind = None;
  std::string gate;
  std::vector<double> angles;
  bool adjoint_flag = false;
  std::vector<c128> matrix;  // row-major, shape given by the owning tensor

  static TensorData none() { return {}; }
  static TensorData from_gate(std::string name, std::vector<double> a = {},
                              bool adjoint = false);
  static TensorData new_from_data(std::vector<c128> data);
  // materialize (tensordata.rs:37-56): flat row-major values
  std::vector<c128> into_data() const;
  TensorData adjoint() const;  // tensordata.rs:59-69
};

// --- gates (gates.rs) ---
// shape is implied: 1q gates 2x2, 2q gates (2,2,2,2); values exact.
std::vector<c128> load_gate(const std::string& name,
                            const std::vector<double>& angles);
std::vector<c128> load_gate_adjoint(const std::string& name,
                                    const std::vector<double>& angles);
// swap first/second half of axes + conjugate over a 2^n x 2^n layout
// (gates.rs:83-101); ndim = number of size-2 axes... general dims supported.
std::vector<c128> matrix_adjoint(const std::vector<c128>& data,
                                 const std::vector<std::uint64_t>& dims);

// --- tensors (tensor.rs) ---
class LeafTensor {
 public:
  LeafTensor() = default;
  LeafTensor(std::vector<EdgeIndex> legs, std::vector<std::uint64_t> dims,
             TensorData data = TensorData::none());
  static LeafTensor new_from_const(std::vector<EdgeIndex> legs,
                                   std::uint64_t dim);

  const std::vector<EdgeIndex>& legs() const { return legs_; }
  const std::vector<std::uint64_t>& bond_dims() const { return dims_; }
  double size() const;  // tensor.rs:571-573
  const TensorData& tensor_data() const { return data_; }
  void set_tensor_data(TensorData d) { data_ = std::move(d); }

  LeafTensor difference(const LeafTensor& o) const;            // tensor.rs:629
  LeafTensor union_with(const LeafTensor& o) const;            // tensor.rs:655
  LeafTensor intersection(const LeafTensor& o) const;          // tensor.rs:683
  LeafTensor symmetric_difference(const LeafTensor& o) const;  // tensor.rs:709

  LeafTensor operator^(const LeafTensor& o) const {
    return symmetric_difference(o);
  }
  LeafTensor operator&(const LeafTensor& o) const { return intersection(o); }
  LeafTensor operator|(const LeafTensor& o) const { return union_with(o); }
  LeafTensor operator-(const LeafTensor& o) const { return difference(o); }

 private:
  std::vector<EdgeIndex> legs_;
  std::vector<std::uint64_t> dims_;
  TensorData data_;
};

// A Tensor is a leaf or a composite (tensor.rs:19-33).
class CompositeTensor;
struct Tensor {
  std::shared_ptr<LeafTensor> leaf;
  std::shared_ptr<CompositeTensor> composite;
  bool is_leaf() const { return leaf != nullptr; }
  Tensor() = default;
  Tensor(LeafTensor t) : leaf(std::make_shared<LeafTensor>(std::move(t))) {}
  Tensor(CompositeTensor t);
};

class CompositeTensor {
 public:
  CompositeTensor() = default;
  explicit CompositeTensor(std::vector<Tensor> tensors)
      : tensors_(std::move(tensors)) {}
  const std::vector<Tensor>& tensors() const { return tensors_; }
  std::vector<Tensor>& tensors() { return tensors_; }
  void push_tensor(Tensor t) { tensors_.push_back(std::move(t)); }
  std::size_t len() const { return tensors_.size(); }
  LeafTensor external_tensor() const;  // tensor.rs:392-403

 private:
  std::vector<Tensor> tensors_;
};

// --- contraction paths (contractionpath.rs) ---
struct ContractionPath {
  std::map<TensorIndex, ContractionPath> nested;
  std::vector<std::pair<TensorIndex, TensorIndex>> toplevel;
  static ContractionPath simple(
      std::vector<std::pair<TensorIndex, TensorIndex>> p) {
    ContractionPath cp;
    cp.toplevel = std::move(p);
    return cp;
  }
};

// SSA -> replace-left (contractionpath.rs:197-215)
ContractionPath ssa_replace_ordering(const ContractionPath& path);

// --- cost model (contraction_cost.rs) ---
double contract_cost_tensors(const LeafTensor& a, const LeafTensor& b);  // :26
double contract_op_cost_tensors(const LeafTensor& a, const LeafTensor& b);
double contract_size_tensors(const LeafTensor& a, const LeafTensor& b);
// (op_cost, mem_cost) over a replace-left path (contraction_cost.rs:101-151)
std::pair<double, double> contract_path_cost(const std::vector<Tensor>& inputs,
                                             const ContractionPath& path,
                                             bool only_count_ops);

// --- pathfinders (paths.rs:21-43, cotengrust.rs) ---
struct ContractionPathResult {
  ContractionPath ssa_path;
  double flops = 0;
  double size = 0;
  ContractionPath replace_path() const { return ssa_replace_ordering(ssa_path); }
};

class Pathfinder {
 public:
  virtual ~Pathfinder() = default;
  virtual ContractionPathResult find_path(const CompositeTensor& tn) = 0;
};

// cotengra-style "memory-removed" greedy (cotengrust.rs Greedy; pinned by
// the reference's expected-path tests, mirrored in tests/host_mirror_test)
class Greedy : public Pathfinder {
 public:
  ContractionPathResult find_path(const CompositeTensor& tn) override;
};

// --- circuit builder (circuit_builder.rs) ---
class Circuit {
 public:
  std::size_t allocate_register(std::size_t n);  // returns base qubit index
  void append_gate(TensorData gate, const std::vector<std::size_t>& qubits);
  std::size_t num_qubits() const { return open_edges_.size(); }
  // bitstring of '0'/'1'/'*'; returns the network (Permutor: the target leg
  // order for the caller to apply)
  std::pair<CompositeTensor, std::vector<EdgeIndex>> into_amplitude_network(
      const std::string& bitstring);
  std::pair<CompositeTensor, std::vector<EdgeIndex>> into_statevector_network();

 private:
  std::vector<EdgeIndex> open_edges_;
  EdgeIndex next_edge_ = 0;
  CompositeTensor tn_;
};

// --- the executor (contraction.rs:35-68), running on the MI355X via the
// tn_net C ABI. Returns the final tensor with data downloaded to host.
LeafTensor contract_tensor_network(const CompositeTensor& tn,
                                   const ContractionPath& replace_path,
                                   int device = 0);

}  // namespace tnc

#endif  // TNC_HOST_HPP
