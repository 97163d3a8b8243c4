#include "align/poa.hpp"

#include <algorithm>
#include <cassert>
#include <limits>
#include <stack>
#include <unordered_set>

namespace rga::poa {

uint32_t Graph::add_node(char letter) {
  nodes_.push_back(Node{letter, {}, {}, {}});
  return static_cast<uint32_t>(nodes_.size() - 1);
}

void Graph::add_edge(uint32_t begin, uint32_t end, int64_t weight) {
  for (uint32_t ei : nodes_[begin].out_edges) {
    if (edges_[ei].end_node == end) {
      edges_[ei].total_weight += weight;
      edges_[ei].labels.emplace_back(num_sequences_);
      return;
    }
  }
  edges_.push_back(Edge{begin, end, weight, {num_sequences_}});
  uint32_t ei = static_cast<uint32_t>(edges_.size() - 1);
  nodes_[begin].out_edges.emplace_back(ei);
  nodes_[end].in_edges.emplace_back(ei);
}

int32_t Graph::add_chain(const char* seq, const std::vector<uint32_t>& weights, uint32_t begin,
                         uint32_t end) {
  if (begin == end) {
    return -1;
  }
  int32_t first_node_id = static_cast<int32_t>(add_node(seq[begin]));
  for (uint32_t i = begin + 1; i < end; ++i) {
    uint32_t node_id = add_node(seq[i]);
    add_edge(node_id - 1, node_id, static_cast<int64_t>(weights[i - 1]) + weights[i]);
  }
  return first_node_id;
}

void Graph::add_alignment(const Alignment& alignment, const char* seq, uint32_t len) {
  std::vector<uint32_t> weights(len, 1);
  add_alignment(alignment, seq, len, weights);
}

void Graph::add_alignment(const Alignment& alignment, const char* seq, uint32_t len,
                          const char* qual, uint32_t qual_len) {
  std::vector<uint32_t> weights;
  weights.reserve(qual_len);
  for (uint32_t i = 0; i < qual_len; ++i) {
    weights.emplace_back(static_cast<uint32_t>(static_cast<uint8_t>(qual[i]) - 33));
  }
  add_alignment(alignment, seq, len, weights);
}

void Graph::add_alignment(const Alignment& alignment, const char* seq, uint32_t len,
                          const std::vector<uint32_t>& weights) {
  if (len == 0) {
    return;
  }
  if (alignment.empty()) {
    add_chain(seq, weights, 0, len);
    ++num_sequences_;
    topological_sort();
    return;
  }

  std::vector<uint32_t> seq_positions;  // aligned (non-gap) sequence positions
  for (const auto& it : alignment) {
    if (it.second != -1) {
      seq_positions.emplace_back(static_cast<uint32_t>(it.second));
    }
  }

  uint32_t nodes_before = static_cast<uint32_t>(nodes_.size());
  add_chain(seq, weights, 0, seq_positions.front());
  int32_t head_node_id =
      nodes_before == nodes_.size() ? -1 : static_cast<int32_t>(nodes_.size() - 1);

  int32_t tail_node_id = add_chain(seq, weights, seq_positions.back() + 1, len);

  int32_t begin_node_id = head_node_id == -1 ? -1 : static_cast<int32_t>(nodes_before);
  int64_t prev_weight = head_node_id == -1 ? 0 : weights[seq_positions.front() - 1];

  for (const auto& it : alignment) {
    if (it.second == -1) {
      continue;
    }
    char letter = seq[it.second];
    int32_t new_node_id;
    if (it.first == -1) {
      new_node_id = static_cast<int32_t>(add_node(letter));
    } else {
      Node& node = nodes_[it.first];
      if (node.letter == letter) {
        new_node_id = it.first;
      } else {
        int32_t aligned_with_same_letter = -1;
        for (uint32_t aid : node.aligned_node_ids) {
          if (nodes_[aid].letter == letter) {
            aligned_with_same_letter = static_cast<int32_t>(aid);
            break;
          }
        }
        if (aligned_with_same_letter == -1) {
          new_node_id = static_cast<int32_t>(add_node(letter));
          for (uint32_t aid : nodes_[it.first].aligned_node_ids) {
            nodes_[new_node_id].aligned_node_ids.emplace_back(aid);
            nodes_[aid].aligned_node_ids.emplace_back(static_cast<uint32_t>(new_node_id));
          }
          nodes_[new_node_id].aligned_node_ids.emplace_back(static_cast<uint32_t>(it.first));
          nodes_[it.first].aligned_node_ids.emplace_back(static_cast<uint32_t>(new_node_id));
        } else {
          new_node_id = aligned_with_same_letter;
        }
      }
    }

    if (begin_node_id == -1) {
      begin_node_id = new_node_id;
    }
    if (head_node_id != -1) {
      add_edge(static_cast<uint32_t>(head_node_id), static_cast<uint32_t>(new_node_id),
               prev_weight + weights[it.second]);
    }
    head_node_id = new_node_id;
    prev_weight = weights[it.second];
  }

  if (tail_node_id != -1) {
    add_edge(static_cast<uint32_t>(head_node_id), static_cast<uint32_t>(tail_node_id),
             prev_weight + weights[seq_positions.back() + 1]);
  }

  ++num_sequences_;
  topological_sort();
}

void Graph::topological_sort() {
  sorted_.clear();
  sorted_.reserve(nodes_.size());

  // 0 - unvisited, 1 - on stack, 2 - done
  std::vector<uint8_t> marks(nodes_.size(), 0);
  // true when this node will be emitted by its aligned-ring leader
  std::vector<bool> emitted_by_leader(nodes_.size(), false);
  std::stack<uint32_t> to_visit;

  for (uint32_t s = 0; s < nodes_.size(); ++s) {
    if (marks[s] != 0) {
      continue;
    }
    to_visit.push(s);
    while (!to_visit.empty()) {
      uint32_t node_id = to_visit.top();
      bool ready = true;
      if (marks[node_id] != 2) {
        for (uint32_t ei : nodes_[node_id].in_edges) {
          if (marks[edges_[ei].begin_node] != 2) {
            to_visit.push(edges_[ei].begin_node);
            ready = false;
          }
        }
        if (!emitted_by_leader[node_id]) {
          for (uint32_t aid : nodes_[node_id].aligned_node_ids) {
            if (marks[aid] != 2) {
              to_visit.push(aid);
              emitted_by_leader[aid] = true;
              ready = false;
            }
          }
        }
        if (ready) {
          marks[node_id] = 2;
          if (!emitted_by_leader[node_id]) {
            sorted_.emplace_back(node_id);
            for (uint32_t aid : nodes_[node_id].aligned_node_ids) {
              sorted_.emplace_back(aid);
            }
          }
        } else {
          marks[node_id] = 1;
        }
      }
      if (ready) {
        to_visit.pop();
      }
    }
  }
  assert(sorted_.size() == nodes_.size());
}

void Graph::traverse_heaviest_bundle() {
  std::vector<int32_t> predecessors(nodes_.size(), -1);
  std::vector<int64_t> scores(nodes_.size(), -1);

  uint32_t max_score_id = 0;
  for (uint32_t node_id : sorted_) {
    for (uint32_t ei : nodes_[node_id].in_edges) {
      const Edge& edge = edges_[ei];
      if (scores[node_id] < edge.total_weight ||
          (scores[node_id] == edge.total_weight && predecessors[node_id] != -1 &&
           scores[predecessors[node_id]] <= scores[edge.begin_node])) {
        scores[node_id] = edge.total_weight;
        predecessors[node_id] = static_cast<int32_t>(edge.begin_node);
      }
    }
    if (predecessors[node_id] != -1) {
      scores[node_id] += scores[predecessors[node_id]];
    }
    if (scores[max_score_id] < scores[node_id]) {
      max_score_id = node_id;
    }
  }

  if (!nodes_[max_score_id].out_edges.empty()) {
    std::vector<uint32_t> rank_of(nodes_.size(), 0);
    for (uint32_t i = 0; i < sorted_.size(); ++i) {
      rank_of[sorted_[i]] = i;
    }
    while (!nodes_[max_score_id].out_edges.empty()) {
      max_score_id = branch_completion(scores, predecessors, rank_of[max_score_id]);
    }
  }

  consensus_.clear();
  int32_t id = static_cast<int32_t>(max_score_id);
  while (predecessors[id] != -1) {
    consensus_.emplace_back(static_cast<uint32_t>(id));
    id = predecessors[id];
  }
  consensus_.emplace_back(static_cast<uint32_t>(id));
  std::reverse(consensus_.begin(), consensus_.end());
}

uint32_t Graph::branch_completion(std::vector<int64_t>& scores,
                                  std::vector<int32_t>& predecessors, uint32_t rank) {
  uint32_t node_id = sorted_[rank];
  // Invalidate the alternatives of this node's outgoing branches so the new
  // search continues strictly through node_id.
  for (uint32_t ei : nodes_[node_id].out_edges) {
    for (uint32_t oe : nodes_[edges_[ei].end_node].in_edges) {
      if (edges_[oe].begin_node != node_id) {
        scores[edges_[oe].begin_node] = -1;
      }
    }
  }

  int64_t max_score = 0;
  uint32_t max_score_id = 0;
  for (uint32_t i = rank + 1; i < sorted_.size(); ++i) {
    uint32_t nid = sorted_[i];
    scores[nid] = -1;
    predecessors[nid] = -1;
    for (uint32_t ei : nodes_[nid].in_edges) {
      const Edge& edge = edges_[ei];
      if (scores[edge.begin_node] == -1) {
        continue;
      }
      if (scores[nid] < edge.total_weight ||
          (scores[nid] == edge.total_weight && predecessors[nid] != -1 &&
           scores[predecessors[nid]] <= scores[edge.begin_node])) {
        scores[nid] = edge.total_weight;
        predecessors[nid] = static_cast<int32_t>(edge.begin_node);
      }
    }
    if (predecessors[nid] != -1) {
      scores[nid] += scores[predecessors[nid]];
    }
    if (max_score < scores[nid]) {
      max_score = scores[nid];
      max_score_id = nid;
    }
  }

  return max_score_id;
}

std::string Graph::generate_consensus(std::vector<uint32_t>* coverages) {
  traverse_heaviest_bundle();

  std::string consensus;
  consensus.reserve(consensus_.size());
  for (uint32_t node_id : consensus_) {
    consensus += nodes_[node_id].letter;
  }

  if (coverages != nullptr) {
    coverages->clear();
    coverages->reserve(consensus_.size());
    std::unordered_set<uint32_t> labels;
    auto add_node_labels = [&](uint32_t nid) {
      for (uint32_t ei : nodes_[nid].in_edges) {
        labels.insert(edges_[ei].labels.begin(), edges_[ei].labels.end());
      }
      for (uint32_t ei : nodes_[nid].out_edges) {
        labels.insert(edges_[ei].labels.begin(), edges_[ei].labels.end());
      }
    };
    for (uint32_t node_id : consensus_) {
      labels.clear();
      add_node_labels(node_id);
      // reads aligned to ring partners cover this consensus column too
      for (uint32_t aid : nodes_[node_id].aligned_node_ids) {
        add_node_labels(aid);
      }
      coverages->emplace_back(static_cast<uint32_t>(labels.size()));
    }
  }
  return consensus;
}

Graph Graph::subgraph(uint32_t begin_node, uint32_t end_node,
                      std::vector<int32_t>* mapping) const {
  // Collect ancestors of end_node (and their aligned rings) with id >= begin.
  std::vector<bool> in_subgraph(nodes_.size(), false);
  std::stack<uint32_t> to_visit;
  to_visit.push(end_node);
  while (!to_visit.empty()) {
    uint32_t node_id = to_visit.top();
    to_visit.pop();
    if (!in_subgraph[node_id] && node_id >= begin_node) {
      for (uint32_t ei : nodes_[node_id].in_edges) {
        to_visit.push(edges_[ei].begin_node);
      }
      for (uint32_t aid : nodes_[node_id].aligned_node_ids) {
        to_visit.push(aid);
      }
      in_subgraph[node_id] = true;
    }
  }

  Graph sub;
  sub.num_sequences_ = num_sequences_;

  mapping->assign(nodes_.size(), -1);  // subgraph id -> parent id (prefix used)
  std::vector<int32_t> parent_to_sub(nodes_.size(), -1);
  for (uint32_t i = 0; i < nodes_.size(); ++i) {
    if (!in_subgraph[i]) {
      continue;
    }
    uint32_t sid = sub.add_node(nodes_[i].letter);
    parent_to_sub[i] = static_cast<int32_t>(sid);
    (*mapping)[sid] = static_cast<int32_t>(i);
  }

  for (uint32_t i = 0; i < nodes_.size(); ++i) {
    if (!in_subgraph[i]) {
      continue;
    }
    uint32_t sid = static_cast<uint32_t>(parent_to_sub[i]);
    for (uint32_t ei : nodes_[i].in_edges) {
      const Edge& edge = edges_[ei];
      if (parent_to_sub[edge.begin_node] != -1) {
        sub.add_edge(static_cast<uint32_t>(parent_to_sub[edge.begin_node]), sid,
                     edge.total_weight);
      }
    }
    for (uint32_t aid : nodes_[i].aligned_node_ids) {
      if (parent_to_sub[aid] != -1) {
        sub.nodes_[sid].aligned_node_ids.emplace_back(static_cast<uint32_t>(parent_to_sub[aid]));
      }
    }
  }

  sub.topological_sort();
  return sub;
}

void Graph::update_alignment(Alignment* alignment, const std::vector<int32_t>& mapping) {
  for (auto& it : *alignment) {
    if (it.first != -1) {
      it.first = mapping[it.first];
    }
  }
}

Alignment NWEngine::align(const char* seq, uint32_t len, const Graph& graph) {
  Alignment alignment;
  if (graph.nodes().empty() || len == 0) {
    return alignment;
  }

  const auto& nodes = graph.nodes();
  const auto& edges = graph.edges();
  const auto& sorted = graph.sorted_node_ids();

  const uint32_t width = len + 1;
  const uint32_t height = static_cast<uint32_t>(nodes.size()) + 1;
  const int32_t kNegInf = std::numeric_limits<int32_t>::min() / 2;

  H_.resize(static_cast<size_t>(width) * height);
  rank_of_.resize(nodes.size());
  for (uint32_t i = 0; i < sorted.size(); ++i) {
    rank_of_[sorted[i]] = i;
  }

  // Row 0: all-gap prefix of the sequence.
  H_[0] = 0;
  for (uint32_t j = 1; j < width; ++j) {
    H_[j] = H_[j - 1] + gap_;
  }
  // Column 0: all-gap prefix of graph paths.
  for (uint32_t r = 0; r < sorted.size(); ++r) {
    const auto& node = nodes[sorted[r]];
    int32_t best_pred = kNegInf;
    if (node.in_edges.empty()) {
      best_pred = 0;
    } else {
      for (uint32_t ei : node.in_edges) {
        best_pred = std::max(best_pred,
                             H_[static_cast<size_t>(rank_of_[edges[ei].begin_node] + 1) * width]);
      }
    }
    H_[static_cast<size_t>(r + 1) * width] = best_pred + gap_;
  }

  int32_t max_score = kNegInf;
  uint32_t max_i = 0;

  for (uint32_t r = 0; r < sorted.size(); ++r) {
    uint32_t node_id = sorted[r];
    const auto& node = nodes[node_id];
    const char letter = node.letter;
    int32_t* H_row = &H_[static_cast<size_t>(r + 1) * width];

    // First predecessor initializes the row; the rest max into it.
    uint32_t pred_i = node.in_edges.empty() ? 0 : rank_of_[edges[node.in_edges[0]].begin_node] + 1;
    {
      const int32_t* H_pred = &H_[static_cast<size_t>(pred_i) * width];
      for (uint32_t j = 1; j < width; ++j) {
        int32_t match = H_pred[j - 1] + (seq[j - 1] == letter ? match_ : mismatch_);
        int32_t del = H_pred[j] + gap_;
        H_row[j] = std::max(match, del);
      }
    }
    for (uint32_t e = 1; e < node.in_edges.size(); ++e) {
      pred_i = rank_of_[edges[node.in_edges[e]].begin_node] + 1;
      const int32_t* H_pred = &H_[static_cast<size_t>(pred_i) * width];
      for (uint32_t j = 1; j < width; ++j) {
        int32_t match = H_pred[j - 1] + (seq[j - 1] == letter ? match_ : mismatch_);
        int32_t del = H_pred[j] + gap_;
        H_row[j] = std::max(H_row[j], std::max(match, del));
      }
    }
    // Horizontal (insertion in sequence) pass.
    for (uint32_t j = 1; j < width; ++j) {
      H_row[j] = std::max(H_row[j], H_row[j - 1] + gap_);
    }

    if (node.out_edges.empty()) {
      if (max_score < H_row[width - 1]) {
        max_score = H_row[width - 1];
        max_i = r + 1;
      }
    }
  }

  // Backtrack from (max_i, len): spoa move priority is diagonal (predecessors
  // in edge order), then vertical/graph gap (same order), then horizontal.
  uint32_t i = max_i, j = width - 1;
  while (!(i == 0 && j == 0)) {
    int32_t H_ij = H_[static_cast<size_t>(i) * width + j];
    uint32_t prev_i = i, prev_j = j;
    bool found = false;

    if (i != 0 && j != 0) {
      const auto& node = nodes[sorted[i - 1]];
      int32_t match_cost = seq[j - 1] == node.letter ? match_ : mismatch_;
      uint32_t pred_i = node.in_edges.empty() ? 0 : rank_of_[edges[node.in_edges[0]].begin_node] + 1;
      if (H_ij == H_[static_cast<size_t>(pred_i) * width + (j - 1)] + match_cost) {
        prev_i = pred_i;
        prev_j = j - 1;
        found = true;
      } else {
        for (uint32_t e = 1; e < node.in_edges.size(); ++e) {
          pred_i = rank_of_[edges[node.in_edges[e]].begin_node] + 1;
          if (H_ij == H_[static_cast<size_t>(pred_i) * width + (j - 1)] + match_cost) {
            prev_i = pred_i;
            prev_j = j - 1;
            found = true;
            break;
          }
        }
      }
    }
    if (!found && i != 0) {
      const auto& node = nodes[sorted[i - 1]];
      uint32_t pred_i = node.in_edges.empty() ? 0 : rank_of_[edges[node.in_edges[0]].begin_node] + 1;
      if (H_ij == H_[static_cast<size_t>(pred_i) * width + j] + gap_) {
        prev_i = pred_i;
        prev_j = j;
        found = true;
      } else {
        for (uint32_t e = 1; e < node.in_edges.size(); ++e) {
          pred_i = rank_of_[edges[node.in_edges[e]].begin_node] + 1;
          if (H_ij == H_[static_cast<size_t>(pred_i) * width + j] + gap_) {
            prev_i = pred_i;
            prev_j = j;
            found = true;
            break;
          }
        }
      }
    }
    if (!found && j != 0) {
      if (H_ij == H_[static_cast<size_t>(i) * width + (j - 1)] + gap_) {
        prev_i = i;
        prev_j = j - 1;
        found = true;
      }
    }
    assert(found && "NWEngine backtrack: no predecessor cell found");

    alignment.emplace_back(i == prev_i ? -1 : static_cast<int32_t>(sorted[i - 1]),
                           j == prev_j ? -1 : static_cast<int32_t>(j - 1));
    i = prev_i;
    j = prev_j;
  }
  std::reverse(alignment.begin(), alignment.end());
  return alignment;
}

}  // namespace rga::poa
