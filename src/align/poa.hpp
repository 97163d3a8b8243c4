// Partial-order-alignment consensus engine.
// Capability parity target: vendor/spoa (v1.1.x line) as called from reference
// src/window.cpp:73-116 and src/polisher.cpp:182-184 — POA graph with
// aligned-node rings, linear-gap Needleman-Wunsch alignment of a sequence to
// the graph (spoa::AlignmentType::kNW), subgraph extraction for partially
// spanning layers, and heaviest-bundle consensus with per-base coverages.
// Reimplemented from the published algorithm (Lee et al. 2002, Vaser et al.
// 2017); tie-breaking chosen to match the reference goldens.
#pragma once

#include <cstdint>
#include <string>
#include <utility>
#include <vector>

namespace rga::poa {

// (node_id, seq_pos) pairs; -1 marks a gap on that side.
using Alignment = std::vector<std::pair<int32_t, int32_t>>;

class Graph {
 public:
  struct Edge {
    uint32_t begin_node;
    uint32_t end_node;
    int64_t total_weight;
    std::vector<uint32_t> labels;  // sequence indices traversing this edge
  };

  struct Node {
    char letter;
    std::vector<uint32_t> in_edges;   // edge indices, creation order
    std::vector<uint32_t> out_edges;  // edge indices, creation order
    std::vector<uint32_t> aligned_node_ids;
  };

  Graph() = default;

  const std::vector<Node>& nodes() const { return nodes_; }
  const std::vector<Edge>& edges() const { return edges_; }
  const std::vector<uint32_t>& sorted_node_ids() const { return sorted_; }

  // Adds `seq` threaded through the graph along `alignment` (empty alignment
  // appends the sequence as a fresh chain). Weights default to 1 per base;
  // the quality overload uses phred (q - 33) weights.
  void add_alignment(const Alignment& alignment, const char* seq, uint32_t len);
  void add_alignment(const Alignment& alignment, const char* seq, uint32_t len, const char* qual,
                     uint32_t qual_len);
  void add_alignment(const Alignment& alignment, const char* seq, uint32_t len,
                     const std::vector<uint32_t>& weights);

  // Heaviest-bundle consensus; coverages[i] = number of sequences supporting
  // consensus base i (used for window-end trimming).
  std::string generate_consensus(std::vector<uint32_t>* coverages);

  // Extracts the subgraph of ancestors of `end_node` with node id >= begin
  // (backbone node ids are 0..len-1, so backbone positions are node ids).
  // mapping: subgraph node id -> parent node id.
  Graph subgraph(uint32_t begin_node, uint32_t end_node, std::vector<int32_t>* mapping) const;

  // Rewrites alignment node ids from subgraph ids to parent ids.
  static void update_alignment(Alignment* alignment, const std::vector<int32_t>& mapping);

 private:
  uint32_t add_node(char letter);
  void add_edge(uint32_t begin, uint32_t end, int64_t weight);
  // Appends seq[begin:end) as a chain; returns first node id or -1 if empty.
  int32_t add_chain(const char* seq, const std::vector<uint32_t>& weights, uint32_t begin,
                    uint32_t end);
  void topological_sort();
  void traverse_heaviest_bundle();
  uint32_t branch_completion(std::vector<int64_t>& scores, std::vector<int32_t>& predecessors,
                             uint32_t rank);

  std::vector<Node> nodes_;
  std::vector<Edge> edges_;
  std::vector<uint32_t> sorted_;
  std::vector<uint32_t> consensus_;
  uint32_t num_sequences_ = 0;
};

// Linear-gap Needleman-Wunsch of a sequence against a POA graph
// (spoa kNW equivalent). One engine per thread; buffers are reused.
class NWEngine {
 public:
  NWEngine(int8_t match, int8_t mismatch, int8_t gap)
      : match_(match), mismatch_(mismatch), gap_(gap) {}

  Alignment align(const char* seq, uint32_t len, const Graph& graph);

 private:
  int8_t match_, mismatch_, gap_;
  std::vector<int32_t> H_;
  std::vector<uint32_t> rank_of_;
};

}  // namespace rga::poa
