#include "core/polisher.hpp"

#include <algorithm>
#include <cstdio>
#include <cstdlib>

#include "align/poa.hpp"

namespace rga {

constexpr uint64_t kChunkSize = 1024ull * 1024 * 1024;  // 1 GiB parse chunks

// Compacts nullptr gaps out of src starting at begin; returns removed count.
template <class T>
static uint64_t compact(std::vector<std::unique_ptr<T>>& src, uint64_t begin) {
  uint64_t i = begin;
  for (uint64_t j = begin; i < src.size(); ++i) {
    if (src[i] != nullptr) {
      continue;
    }
    j = std::max(j, i);
    while (j < src.size() && src[j] == nullptr) {
      ++j;
    }
    if (j >= src.size()) {
      break;
    }
    if (i != j) {
      src[i].swap(src[j]);
    }
  }
  uint64_t removed = src.size() - i;
  if (i < src.size()) {
    src.resize(i);
  }
  return removed;
}

std::unique_ptr<Polisher> createPolisher(const std::string& sequences_path,
                                         const std::string& overlaps_path,
                                         const std::string& target_path, PolisherConfig config) {
  if (config.type != PolisherType::kC && config.type != PolisherType::kF) {
    fprintf(stderr, "[racon::createPolisher] error: invalid polisher type!\n");
    exit(1);
  }
  if (config.window_length == 0) {
    fprintf(stderr, "[racon::createPolisher] error: invalid window length!\n");
    exit(1);
  }

  auto sparser = createSequenceParser(sequences_path);
  auto oparser = createOverlapParser(overlaps_path);
  auto tparser = createSequenceParser(target_path);

  if (config.poa_batches > 0 || config.aligner_batches > 0) {
    extern std::unique_ptr<Polisher> createHipPolisher(std::unique_ptr<SequenceParser>,
                                                       std::unique_ptr<OverlapParser>,
                                                       std::unique_ptr<SequenceParser>,
                                                       PolisherConfig);
    return createHipPolisher(std::move(sparser), std::move(oparser), std::move(tparser), config);
  }
  return std::make_unique<Polisher>(std::move(sparser), std::move(oparser), std::move(tparser),
                                    config);
}

Polisher::Polisher(std::unique_ptr<SequenceParser> sparser, std::unique_ptr<OverlapParser> oparser,
                   std::unique_ptr<SequenceParser> tparser, PolisherConfig config)
    : sparser_(std::move(sparser)),
      oparser_(std::move(oparser)),
      tparser_(std::move(tparser)),
      config_(config),
      dummy_quality_(config.window_length, '!'),
      thread_pool_(std::make_unique<ThreadPool>(config.num_threads)),
      logger_(std::make_unique<Logger>()) {
  for (uint32_t i = 0; i < thread_pool_->num_threads(); ++i) {
    engines_.emplace_back(
        std::make_unique<poa::NWEngine>(config.match, config.mismatch, config.gap));
  }
}

Polisher::~Polisher() {
  if (logger_) {
    logger_->total("[racon::Polisher::] total =");
  }
}

void Polisher::initialize() {
  if (!windows_.empty()) {
    fprintf(stderr, "[racon::Polisher::initialize] warning: object already initialized!\n");
    return;
  }

  logger_->log();

  tparser_->reset();
  tparser_->parse(sequences_, static_cast<uint64_t>(-1));

  uint64_t targets_size = sequences_.size();
  if (targets_size == 0) {
    fprintf(stderr, "[racon::Polisher::initialize] error: empty target sequences set!\n");
    exit(1);
  }

  std::unordered_map<std::string, uint64_t> name_to_id;
  std::unordered_map<uint64_t, uint64_t> id_to_id;
  for (uint64_t i = 0; i < targets_size; ++i) {
    name_to_id[sequences_[i]->name() + "t"] = i;
    id_to_id[i << 1 | 1] = i;
  }

  std::vector<bool> has_name(targets_size, true);
  std::vector<bool> has_data(targets_size, true);
  std::vector<bool> has_reverse_data(targets_size, false);

  logger_->log("[racon::Polisher::initialize] loaded target sequences");
  logger_->log();

  uint64_t sequences_size = 0, total_sequences_length = 0;

  sparser_->reset();
  while (true) {
    uint64_t l = sequences_.size();
    bool more = sparser_->parse(sequences_, kChunkSize);

    uint64_t n = 0;
    for (uint64_t i = l; i < sequences_.size(); ++i, ++sequences_size) {
      total_sequences_length += sequences_[i]->data().size();

      auto it = name_to_id.find(sequences_[i]->name() + "t");
      if (it != name_to_id.end()) {
        if (sequences_[i]->data().size() != sequences_[it->second]->data().size() ||
            sequences_[i]->quality().size() != sequences_[it->second]->quality().size()) {
          fprintf(stderr,
                  "[racon::Polisher::initialize] error: duplicate sequence %s with unequal data\n",
                  sequences_[i]->name().c_str());
          exit(1);
        }
        name_to_id[sequences_[i]->name() + "q"] = it->second;
        id_to_id[sequences_size << 1 | 0] = it->second;
        sequences_[i].reset();
        ++n;
      } else {
        name_to_id[sequences_[i]->name() + "q"] = i - n;
        id_to_id[sequences_size << 1 | 0] = i - n;
      }
    }

    compact(sequences_, l);
    if (!more) {
      break;
    }
  }

  if (sequences_size == 0) {
    fprintf(stderr, "[racon::Polisher::initialize] error: empty sequences set!\n");
    exit(1);
  }

  has_name.resize(sequences_.size(), false);
  has_data.resize(sequences_.size(), false);
  has_reverse_data.resize(sequences_.size(), false);

  window_type_ = static_cast<double>(total_sequences_length) / sequences_size <= 1000
                     ? WindowType::kNGS
                     : WindowType::kTGS;

  logger_->log("[racon::Polisher::initialize] loaded sequences");
  logger_->log();

  std::vector<std::unique_ptr<Overlap>> overlaps;

  auto remove_invalid_overlaps = [&](uint64_t begin, uint64_t end) {
    for (uint64_t i = begin; i < end; ++i) {
      if (overlaps[i] == nullptr) {
        continue;
      }
      if (overlaps[i]->error() > config_.error_threshold ||
          overlaps[i]->q_id() == overlaps[i]->t_id()) {
        overlaps[i].reset();
        continue;
      }
      if (config_.type == PolisherType::kC) {
        // keep only the longest overlap per query
        for (uint64_t j = i + 1; j < end; ++j) {
          if (overlaps[j] == nullptr) {
            continue;
          }
          if (overlaps[i]->length() > overlaps[j]->length()) {
            overlaps[j].reset();
          } else {
            overlaps[i].reset();
            break;
          }
        }
      }
    }
  };

  oparser_->reset();
  uint64_t l = 0;
  while (true) {
    bool more = oparser_->parse(overlaps, kChunkSize);

    uint64_t c = l;
    for (uint64_t i = l; i < overlaps.size(); ++i) {
      overlaps[i]->resolve_ids(sequences_, name_to_id, id_to_id);
      if (!overlaps[i]->is_valid()) {
        overlaps[i].reset();
        continue;
      }
      while (overlaps[c] == nullptr) {
        ++c;
      }
      if (overlaps[c]->q_id() != overlaps[i]->q_id()) {
        remove_invalid_overlaps(c, i);
        c = i;
      }
    }
    if (!more) {
      remove_invalid_overlaps(c, overlaps.size());
      c = overlaps.size();
    }

    for (uint64_t i = l; i < c; ++i) {
      if (overlaps[i] == nullptr) {
        continue;
      }
      if (overlaps[i]->strand()) {
        has_reverse_data[overlaps[i]->q_id()] = true;
      } else {
        has_data[overlaps[i]->q_id()] = true;
      }
    }

    uint64_t removed = compact(overlaps, l);
    l = c - removed;
    if (!more) {
      break;
    }
  }

  std::unordered_map<std::string, uint64_t>().swap(name_to_id);
  std::unordered_map<uint64_t, uint64_t>().swap(id_to_id);

  if (overlaps.empty()) {
    fprintf(stderr, "[racon::Polisher::initialize] error: empty overlap set!\n");
    exit(1);
  }

  logger_->log("[racon::Polisher::initialize] loaded overlaps");
  logger_->log();

  {
    std::vector<std::future<void>> futures;
    for (uint64_t i = 0; i < sequences_.size(); ++i) {
      futures.emplace_back(thread_pool_->submit(
          [&](uint64_t j) {
            sequences_[j]->release(has_name[j], has_data[j], has_reverse_data[j]);
          },
          i));
    }
    for (const auto& f : futures) {
      f.wait();
    }
  }

  find_overlap_breaking_points(overlaps);

  logger_->log();

  std::vector<uint64_t> id_to_first_window_id(targets_size + 1, 0);
  for (uint64_t i = 0; i < targets_size; ++i) {
    uint32_t k = 0;
    for (uint32_t j = 0; j < sequences_[i]->data().size(); j += config_.window_length, ++k) {
      uint32_t length = std::min(j + config_.window_length,
                                 static_cast<uint32_t>(sequences_[i]->data().size())) -
                        j;
      windows_.emplace_back(createWindow(
          i, k, window_type_, &(sequences_[i]->data()[j]), length,
          sequences_[i]->quality().empty() ? &(dummy_quality_[0]) : &(sequences_[i]->quality()[j]),
          length));
    }
    id_to_first_window_id[i + 1] = id_to_first_window_id[i] + k;
  }

  targets_coverages_.assign(targets_size, 0);

  for (uint64_t i = 0; i < overlaps.size(); ++i) {
    ++targets_coverages_[overlaps[i]->t_id()];

    const auto& sequence = sequences_[overlaps[i]->q_id()];
    const auto& breaking_points = overlaps[i]->breaking_points();

    for (uint32_t j = 0; j < breaking_points.size(); j += 2) {
      if (breaking_points[j + 1].second - breaking_points[j].second <
          0.02 * config_.window_length) {
        continue;
      }

      if (!sequence->quality().empty() || !sequence->reverse_quality().empty()) {
        const auto& quality =
            overlaps[i]->strand() ? sequence->reverse_quality() : sequence->quality();
        double average_quality = 0;
        for (uint32_t k = breaking_points[j].second; k < breaking_points[j + 1].second; ++k) {
          average_quality += static_cast<uint32_t>(quality[k]) - 33;
        }
        average_quality /= breaking_points[j + 1].second - breaking_points[j].second;
        if (average_quality < config_.quality_threshold) {
          continue;
        }
      }

      uint64_t window_id = id_to_first_window_id[overlaps[i]->t_id()] +
                           breaking_points[j].first / config_.window_length;
      uint32_t window_start =
          (breaking_points[j].first / config_.window_length) * config_.window_length;

      const char* data = overlaps[i]->strand()
                             ? &(sequence->reverse_complement()[breaking_points[j].second])
                             : &(sequence->data()[breaking_points[j].second]);
      uint32_t data_length = breaking_points[j + 1].second - breaking_points[j].second;

      const char* quality =
          overlaps[i]->strand()
              ? (sequence->reverse_quality().empty()
                     ? nullptr
                     : &(sequence->reverse_quality()[breaking_points[j].second]))
              : (sequence->quality().empty() ? nullptr
                                             : &(sequence->quality()[breaking_points[j].second]));
      uint32_t quality_length = quality == nullptr ? 0 : data_length;

      windows_[window_id]->add_layer(data, data_length, quality, quality_length,
                                     breaking_points[j].first - window_start,
                                     breaking_points[j + 1].first - window_start - 1);
    }

    overlaps[i].reset();
  }

  logger_->log("[racon::Polisher::initialize] transformed data into windows");
}

void Polisher::find_overlap_breaking_points(std::vector<std::unique_ptr<Overlap>>& overlaps) {
  std::vector<std::future<void>> futures;
  futures.reserve(overlaps.size());
  for (uint64_t i = 0; i < overlaps.size(); ++i) {
    futures.emplace_back(thread_pool_->submit(
        [&](uint64_t j) { overlaps[j]->find_breaking_points(sequences_, config_.window_length); },
        i));
  }

  uint64_t logger_step = futures.size() / 20;
  for (uint64_t i = 0; i < futures.size(); ++i) {
    futures[i].wait();
    if (logger_step != 0 && (i + 1) % logger_step == 0 && (i + 1) / logger_step < 20) {
      logger_->bar("[racon::Polisher::initialize] aligning overlaps");
    }
  }
  if (logger_step != 0) {
    logger_->bar("[racon::Polisher::initialize] aligning overlaps");
  } else {
    logger_->log("[racon::Polisher::initialize] aligned overlaps");
  }
}

void Polisher::generate_consensus_cpu(std::vector<bool>& polished,
                                      const std::vector<bool>* todo) {
  polished.resize(windows_.size(), false);
  std::vector<std::future<std::pair<uint64_t, bool>>> futures;
  for (uint64_t i = 0; i < windows_.size(); ++i) {
    if (todo != nullptr && !(*todo)[i]) {
      continue;
    }
    futures.emplace_back(thread_pool_->submit(
        [&](uint64_t j) -> std::pair<uint64_t, bool> {
          uint32_t tid = thread_pool_->this_thread_id();
          if (tid == ~0u) {
            fprintf(stderr, "[racon::Polisher::polish] error: thread identifier not present!\n");
            exit(1);
          }
          return {j, windows_[j]->generate_consensus(*engines_[tid], config_.trim)};
        },
        i));
  }

  uint64_t logger_step = futures.size() / 20;
  for (uint64_t i = 0; i < futures.size(); ++i) {
    auto result = futures[i].get();
    polished[result.first] = result.second;
    if (logger_step != 0 && (i + 1) % logger_step == 0 && (i + 1) / logger_step < 20) {
      logger_->bar("[racon::Polisher::polish] generating consensus");
    }
  }
  if (logger_step != 0) {
    logger_->bar("[racon::Polisher::polish] generating consensus");
  } else {
    logger_->log("[racon::Polisher::polish] generated consensus");
  }
}

void Polisher::collect(std::vector<std::unique_ptr<Sequence>>& dst, bool drop_unpolished,
                       const std::vector<bool>& polished) {
  std::string polished_data;
  uint32_t num_polished_windows = 0;

  for (uint64_t i = 0; i < windows_.size(); ++i) {
    num_polished_windows += polished[i] ? 1 : 0;
    polished_data += windows_[i]->consensus();

    if (i == windows_.size() - 1 || windows_[i + 1]->rank() == 0) {
      double polished_ratio =
          num_polished_windows / static_cast<double>(windows_[i]->rank() + 1);

      if (!drop_unpolished || polished_ratio > 0) {
        std::string tags = config_.type == PolisherType::kF ? "r" : "";
        tags += " LN:i:" + std::to_string(polished_data.size());
        tags += " RC:i:" + std::to_string(targets_coverages_[windows_[i]->id()]);
        tags += " XC:f:" + std::to_string(polished_ratio);
        dst.emplace_back(createSequence(sequences_[windows_[i]->id()]->name() + tags,
                                        polished_data));
      }
      num_polished_windows = 0;
      polished_data.clear();
    }
    windows_[i].reset();
  }

  std::vector<std::shared_ptr<Window>>().swap(windows_);
  std::vector<std::unique_ptr<Sequence>>().swap(sequences_);
}

void Polisher::polish(std::vector<std::unique_ptr<Sequence>>& dst, bool drop_unpolished) {
  logger_->log();
  std::vector<bool> polished;
  generate_consensus_cpu(polished, nullptr);
  collect(dst, drop_unpolished, polished);
}

}  // namespace rga
