// bflc_amd._ledger — deterministic committee-consensus state machine.
//
// Re-implements the semantics of the reference on-chain coordinator
// (reference: FISCO-BCOS/libprecompiled/extension/CommitteePrecompiled.cpp,
// functions RegisterNode/QueryState/QueryGlobalModel/UploadLocalUpdate/
// UploadScores/QueryAllUpdates, .cpp:132-311, and Aggregate .cpp:349-455)
// as a plain deterministic C++ library: no chain, no tables — every rank
// holds a replica and feeds it the same submissions in the same order
// (rank-ordered all-gather), which yields identical state on all ranks.
//
// The numeric aggregation (weighted FedAvg over the selected deltas) is
// NOT done here: upload_scores() returns an AggregationDecision (selected
// trainers + weights + new roles) and the caller applies it with the
// weighted-reduce HIP kernel, then calls commit_aggregate(). This keeps
// the ledger pure/deterministic and the O(model) math on the GPU.
//
// Deliberate divergences from the reference (documented in DESIGN.md):
//  - initial committee = first comm_count registrants in registration
//    order (reference: unordered_map iteration order, .cpp:175-186);
//  - ties in top-k broken by (score desc, id asc) (reference: unstable
//    std::sort with cmp_by_value, .cpp:118-120, 364-366);
//  - duplicate score upload overwrites without double-counting
//    (reference increments score_count unconditionally, .cpp:279-289);
//  - a committee rotation that found fewer scored trainers than
//    comm_count refills the remaining seats in registration order
//    (the reference's committee shrinks while its aggregation trigger
//    stays at COMM_COUNT — a permanent deadlock; found by the
//    hypothesis fuzz in tests/test_ledger_property.py).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cstdlib>
#include <iostream>
#include <map>
#include <optional>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

namespace bflc {

// Admission result codes for upload_local_update (mirrors the silent
// early-returns of CommitteePrecompiled.cpp:225-244).
enum class Admit : int {
  kAccepted = 0,
  kStaleEpoch = 1,
  kDuplicate = 2,
  kQuotaFull = 3,
  kNotStarted = 4,  // epoch still -999 (FL not started)
  kFinished = 5,    // epoch > max_epoch (reference main.py:251-252)
  kBadSignature = 6,  // origin authentication failed (chain layer)
};

struct LedgerConfig {
  int client_num = 20;         // reference CommitteePrecompiled.h:17
  int comm_count = 4;          // .h:11
  int needed_update_count = 10;  // .h:15
  int aggregate_count = 6;     // .h:13
  double learning_rate = 1e-3;  // .h:19 (applied by the engine, not here)
  int max_epoch = 1000;        // main.py:65
};

struct Update {
  std::string blob;  // opaque serialized delta (JSON or binary tensor bytes)
  long n_samples = 0;
  double avg_cost = 0.0;
};

struct AggregationDecision {
  int epoch = 0;  // the epoch being aggregated
  // (trainer id, weight=n_samples) in aggregation order — deterministic.
  std::vector<std::pair<std::string, long>> selected;
  double total_weight = 0.0;
  double avg_cost = 0.0;  // mean of selected trainers' avg_cost
  std::vector<std::string> next_committee;  // top comm_count scored trainers
  std::map<std::string, double> median_scores;  // per-trainer medians
};

// Protocol trace, the reference's OUTPUT-gated log lines
// (CommitteePrecompiled.h:4, .cpp:240-243,255-257,291-293,422-425):
// BFLC_LEDGER_TRACE=1 prints the same progress markers to stderr so an
// operator can eyeball replicas the way the reference README documents
// (tail-ing 4 chain-node logs).
inline bool ledger_trace() {
  static const bool on = [] {
    const char* e = std::getenv("BFLC_LEDGER_TRACE");
    return e && std::atoi(e) != 0;
  }();
  return on;
}

// Median with the exact semantics of GetMid (CommitteePrecompiled.cpp:81-115):
// sorted ascending, odd n -> v[n/2]; even n -> (v[n/2-1] + v[n/2]) / 2.
// The reference computes in float; we match that rounding.
inline float median_ref(std::vector<float> v) {
  if (v.empty()) throw std::invalid_argument("median of empty vector");
  std::sort(v.begin(), v.end());
  const size_t n = v.size();
  if (n % 2 == 1) return v[n / 2];
  return (v[n / 2 - 1] + v[n / 2]) / 2.0f;
}

class CommitteeLedger {
 public:
  explicit CommitteeLedger(LedgerConfig cfg) : cfg_(cfg) {
    if (cfg_.comm_count < 1 || cfg_.client_num < 1)
      throw std::invalid_argument("comm_count and client_num must be >= 1");
    if (cfg_.aggregate_count > cfg_.needed_update_count)
      throw std::invalid_argument("aggregate_count > needed_update_count");
  }

  const LedgerConfig& config() const { return cfg_; }

  // --- RegisterNode (reference .cpp:168-190) -----------------------------
  // Returns true if this registration was new. When the client_num-th node
  // registers, the first comm_count registrants become the committee and
  // epoch goes -999 -> 0.
  bool register_node(const std::string& origin) {
    if (roles_.count(origin)) return false;
    roles_[origin] = "trainer";
    reg_order_.push_back(origin);
    if (static_cast<int>(roles_.size()) == cfg_.client_num) {
      for (int i = 0; i < cfg_.comm_count &&
                      i < static_cast<int>(reg_order_.size());
           ++i)
        roles_[reg_order_[i]] = "comm";
      epoch_ = 0;
    }
    return true;
  }

  // --- QueryState (reference .cpp:191-206) --------------------------------
  // Unregistered nodes read back "trainer" without being persisted.
  std::pair<std::string, int> query_state(const std::string& origin) const {
    auto it = roles_.find(origin);
    return {it == roles_.end() ? std::string("trainer") : it->second, epoch_};
  }

  // --- QueryGlobalModel (reference .cpp:207-214) --------------------------
  std::pair<py::bytes, int> query_global_model() const {
    return {py::bytes(global_model_), epoch_};
  }

  void set_global_model(const std::string& blob) { global_model_ = blob; }

  // --- UploadLocalUpdate (reference .cpp:215-257) -------------------------
  Admit upload_local_update(const std::string& origin, const std::string& blob,
                            int epoch, long n_samples, double avg_cost) {
    if (epoch_ <= kEpochUninit) return Admit::kNotStarted;
    if (finished()) return Admit::kFinished;
    if (epoch != epoch_) return Admit::kStaleEpoch;
    if (local_updates_.count(origin)) return Admit::kDuplicate;
    if (update_count_ >= cfg_.needed_update_count) return Admit::kQuotaFull;
    ++update_count_;
    local_updates_[origin] = Update{blob, n_samples, avg_cost};
    update_order_.push_back(origin);
    if (ledger_trace())
      std::clog << "the update of local model is collected! ("
                << update_count_ << "/" << cfg_.needed_update_count
                << ", epoch " << epoch_ << ", " << origin << ")\n";
    return Admit::kAccepted;
  }

  // --- QueryAllUpdates (reference .cpp:299-311) ---------------------------
  // Empty until the quota is reached (the committee re-polls), then the
  // full map. Returned in acceptance order.
  std::vector<std::pair<std::string, py::bytes>> query_all_updates() const {
    std::vector<std::pair<std::string, py::bytes>> out;
    if (update_count_ < cfg_.needed_update_count) return out;
    for (const auto& id : update_order_)
      out.emplace_back(id, py::bytes(local_updates_.at(id).blob));
    return out;
  }

  bool updates_ready() const {
    return update_count_ >= cfg_.needed_update_count;
  }

  // --- UploadScores (reference .cpp:259-297) ------------------------------
  // Returns the aggregation decision when this was the comm_count-th score
  // set, std::nullopt otherwise (including on rejected submissions).
  //
  // Hostile-input hardening (round 2): score keys that name no ADMITTED
  // update are dropped at admission — the reference only ever reads the
  // score maps for keys that exist in its updates table (.cpp:351-377),
  // so a garbage key there is inert; here it must not survive into
  // decide_aggregation where it could differ across replicas or throw.
  std::optional<AggregationDecision> upload_scores(
      const std::string& origin, int epoch,
      const std::map<std::string, double>& scores) {
    if (pending_) throw std::logic_error("aggregation pending, commit first");
    if (epoch_ <= kEpochUninit || finished() || epoch != epoch_)
      return std::nullopt;
    auto it = roles_.find(origin);
    if (it == roles_.end() || it->second == "trainer") return std::nullopt;
    std::map<std::string, double> filtered;
    for (const auto& kv : scores)
      if (local_updates_.count(kv.first)) filtered.emplace(kv);
    const bool existed = local_scores_.count(origin) > 0;
    local_scores_[origin] = std::move(filtered);
    if (!existed) ++score_count_;
    if (ledger_trace())
      std::clog << score_count_ << " scores has been uploaded (epoch "
                << epoch_ << ")\n";
    if (score_count_ == cfg_.comm_count) {
      pending_ = decide_aggregation();
      return pending_;
    }
    return std::nullopt;
  }

  // --- Aggregate commit (reference .cpp:403-455, state-mutating half) ----
  // The engine computed: new_global = old_global - lr * weighted_avg_delta
  // (CommitteePrecompiled.cpp:403-414) and hands the new blob back.
  void commit_aggregate(const std::string& new_global_blob) {
    if (!pending_) throw std::logic_error("no pending aggregation");
    global_model_ = new_global_blob;
    global_loss_ = pending_->avg_cost;
    if (ledger_trace())
      std::clog << "the " << epoch_ << " epoch , global loss : "
                << global_loss_ << "\n";
    epoch_ += 1;
    local_updates_.clear();
    update_order_.clear();
    local_scores_.clear();
    update_count_ = 0;
    score_count_ = 0;
    // Role rotation (.cpp:443-455): all comm -> trainer, then the top
    // comm_count scored trainers -> comm. Divergence from the
    // reference: if fewer trainers were scored than comm_count, the
    // reference's committee SHRINKS while the score_count==COMM_COUNT
    // aggregation trigger stays — a permanent deadlock (latent, masked
    // there by 16 trainers racing for 10 slots). Here the remaining
    // seats are refilled deterministically in registration order, so
    // the protocol is live for any admitted-update count >= 1.
    for (auto& kv : roles_)
      if (kv.second == "comm") kv.second = "trainer";
    int seated = 0;
    for (const auto& id : pending_->next_committee) {
      roles_[id] = "comm";
      ++seated;
    }
    for (const auto& id : reg_order_) {
      if (seated >= cfg_.comm_count) break;
      if (roles_[id] != "comm") {
        roles_[id] = "comm";
        ++seated;
      }
    }
    pending_.reset();
  }

  // --- introspection ------------------------------------------------------
  int epoch() const { return epoch_; }
  // The FL run is over once epoch exceeds max_epoch (reference clients
  // exit on epoch > MAX_EPOCH, main.py:251-252); the ledger refuses
  // further uploads so no replica can be driven past the end.
  bool finished() const {
    return epoch_ > kEpochUninit && epoch_ > cfg_.max_epoch;
  }
  int update_count() const { return update_count_; }
  int score_count() const { return score_count_; }
  double global_loss() const { return global_loss_; }
  std::map<std::string, std::string> roles() const { return roles_; }
  std::vector<std::string> registration_order() const { return reg_order_; }
  std::vector<std::string> committee() const {
    std::vector<std::string> out;
    for (const auto& kv : roles_)
      if (kv.second == "comm") out.push_back(kv.first);
    return out;
  }
  std::vector<std::string> trainers() const {
    std::vector<std::string> out;
    for (const auto& kv : roles_)
      if (kv.second == "trainer") out.push_back(kv.first);
    return out;
  }
  std::map<std::string, std::map<std::string, double>> local_scores() const {
    return local_scores_;
  }
  py::bytes update_blob(const std::string& id) const {
    return py::bytes(local_updates_.at(id).blob);
  }
  std::pair<long, double> update_meta(const std::string& id) const {
    const auto& u = local_updates_.at(id);
    return {u.n_samples, u.avg_cost};
  }
  bool has_pending() const { return pending_.has_value(); }

  // --- checkpoint/resume ---------------------------------
  // Full-state snapshot as a python dict; restore() rebuilds it. The
  // on-disk JSON serialization lives in the Python facade.
  py::dict snapshot() const {
    py::dict d;
    d["epoch"] = epoch_;
    d["global_model"] = py::bytes(global_model_);
    d["global_loss"] = global_loss_;
    d["update_count"] = update_count_;
    d["score_count"] = score_count_;
    d["roles"] = roles_;
    d["reg_order"] = reg_order_;
    d["update_order"] = update_order_;
    py::dict ups;
    for (const auto& kv : local_updates_) {
      py::dict u;
      u["blob"] = py::bytes(kv.second.blob);
      u["n_samples"] = kv.second.n_samples;
      u["avg_cost"] = kv.second.avg_cost;
      ups[py::str(kv.first)] = u;
    }
    d["local_updates"] = ups;
    d["local_scores"] = local_scores_;
    return d;
  }

  void restore(const py::dict& d) {
    epoch_ = d["epoch"].cast<int>();
    global_model_ = d["global_model"].cast<std::string>();
    global_loss_ = d["global_loss"].cast<double>();
    update_count_ = d["update_count"].cast<int>();
    score_count_ = d["score_count"].cast<int>();
    roles_ = d["roles"].cast<std::map<std::string, std::string>>();
    reg_order_ = d["reg_order"].cast<std::vector<std::string>>();
    update_order_ = d["update_order"].cast<std::vector<std::string>>();
    local_updates_.clear();
    for (auto item : d["local_updates"].cast<py::dict>()) {
      auto u = item.second.cast<py::dict>();
      local_updates_[item.first.cast<std::string>()] =
          Update{u["blob"].cast<std::string>(), u["n_samples"].cast<long>(),
                 u["avg_cost"].cast<double>()};
    }
    local_scores_ =
        d["local_scores"]
            .cast<std::map<std::string, std::map<std::string, double>>>();
    pending_.reset();
  }

 private:
  static constexpr int kEpochUninit = -999;  // reference .cpp:322

  // Median -> sort -> top-k (reference .cpp:349-400 decision half).
  AggregationDecision decide_aggregation() const {
    // Per-trainer median over the committee score sets. Exactly the
    // ADMITTED updates participate (reference iterates its updates
    // table, .cpp:351-362); a committee member that omitted an admitted
    // trainer contributes 0.0 for it — so every admitted trainer always
    // has score_count values and decide can never throw, whatever a
    // hostile/buggy committee member uploaded.
    AggregationDecision dec;
    dec.epoch = epoch_;
    std::vector<std::pair<std::string, float>> ranked;
    for (const auto& id : update_order_) {
      std::vector<float> vals;
      vals.reserve(local_scores_.size());
      for (const auto& comm_kv : local_scores_) {
        auto sit = comm_kv.second.find(id);
        vals.push_back(sit == comm_kv.second.end()
                           ? 0.0f
                           : static_cast<float>(sit->second));
      }
      float m = vals.empty() ? 0.0f : median_ref(vals);
      dec.median_scores[id] = m;
      ranked.emplace_back(id, m);
    }
    // Deterministic: score desc, then id asc (reference: unstable sort).
    std::sort(ranked.begin(), ranked.end(), [](const auto& a, const auto& b) {
      if (a.second != b.second) return a.second > b.second;
      return a.first < b.first;
    });

    const int k =
        std::min<int>(cfg_.aggregate_count, static_cast<int>(ranked.size()));
    for (int i = 0; i < k; ++i) {
      const auto& id = ranked[i].first;
      // ranked is built from update_order_, so the lookup always hits;
      // keep the skip (not a throw) as a restore()-path safety net.
      auto it = local_updates_.find(id);
      if (it == local_updates_.end()) continue;
      dec.selected.emplace_back(id, it->second.n_samples);
      dec.total_weight += static_cast<double>(it->second.n_samples);
      dec.avg_cost += it->second.avg_cost;
    }
    if (!dec.selected.empty())
      dec.avg_cost /= static_cast<double>(dec.selected.size());
    const int c =
        std::min<int>(cfg_.comm_count, static_cast<int>(ranked.size()));
    for (int i = 0; i < c; ++i) dec.next_committee.push_back(ranked[i].first);
    return dec;
  }

  LedgerConfig cfg_;
  int epoch_ = kEpochUninit;
  std::string global_model_;
  double global_loss_ = 0.0;
  std::map<std::string, std::string> roles_;
  std::vector<std::string> reg_order_;
  std::map<std::string, Update> local_updates_;
  std::vector<std::string> update_order_;
  std::map<std::string, std::map<std::string, double>> local_scores_;
  int update_count_ = 0;
  int score_count_ = 0;
  std::optional<AggregationDecision> pending_;
};

}  // namespace bflc

PYBIND11_MODULE(_ledger, m) {
  m.doc() = "bflc_amd deterministic committee ledger (C++)";

  py::enum_<bflc::Admit>(m, "Admit")
      .value("ACCEPTED", bflc::Admit::kAccepted)
      .value("STALE_EPOCH", bflc::Admit::kStaleEpoch)
      .value("DUPLICATE", bflc::Admit::kDuplicate)
      .value("QUOTA_FULL", bflc::Admit::kQuotaFull)
      .value("NOT_STARTED", bflc::Admit::kNotStarted)
      .value("FINISHED", bflc::Admit::kFinished)
      .value("BAD_SIGNATURE", bflc::Admit::kBadSignature);

  py::class_<bflc::LedgerConfig>(m, "LedgerConfig")
      .def(py::init<>())
      .def_readwrite("client_num", &bflc::LedgerConfig::client_num)
      .def_readwrite("comm_count", &bflc::LedgerConfig::comm_count)
      .def_readwrite("needed_update_count",
                     &bflc::LedgerConfig::needed_update_count)
      .def_readwrite("aggregate_count", &bflc::LedgerConfig::aggregate_count)
      .def_readwrite("learning_rate", &bflc::LedgerConfig::learning_rate)
      .def_readwrite("max_epoch", &bflc::LedgerConfig::max_epoch);

  py::class_<bflc::AggregationDecision>(m, "AggregationDecision")
      .def_readonly("epoch", &bflc::AggregationDecision::epoch)
      .def_readonly("selected", &bflc::AggregationDecision::selected)
      .def_readonly("total_weight", &bflc::AggregationDecision::total_weight)
      .def_readonly("avg_cost", &bflc::AggregationDecision::avg_cost)
      .def_readonly("next_committee",
                    &bflc::AggregationDecision::next_committee)
      .def_readonly("median_scores",
                    &bflc::AggregationDecision::median_scores);

  m.def("median_ref", [](std::vector<float> v) { return bflc::median_ref(v); },
        "Median with the reference GetMid semantics "
        "(CommitteePrecompiled.cpp:81-115)");

  py::class_<bflc::CommitteeLedger>(m, "CommitteeLedger")
      .def(py::init<bflc::LedgerConfig>())
      .def_property_readonly("config", &bflc::CommitteeLedger::config)
      .def("register_node", &bflc::CommitteeLedger::register_node)
      .def("query_state", &bflc::CommitteeLedger::query_state)
      .def("query_global_model", &bflc::CommitteeLedger::query_global_model)
      .def("set_global_model",
           [](bflc::CommitteeLedger& l, py::bytes b) {
             l.set_global_model(std::string(b));
           })
      .def("upload_local_update",
           [](bflc::CommitteeLedger& l, const std::string& origin,
              py::bytes blob, int epoch, long n_samples, double avg_cost) {
             return l.upload_local_update(origin, std::string(blob), epoch,
                                          n_samples, avg_cost);
           })
      .def("query_all_updates", &bflc::CommitteeLedger::query_all_updates)
      .def("updates_ready", &bflc::CommitteeLedger::updates_ready)
      .def("upload_scores", &bflc::CommitteeLedger::upload_scores)
      .def("commit_aggregate",
           [](bflc::CommitteeLedger& l, py::bytes b) {
             l.commit_aggregate(std::string(b));
           })
      .def_property_readonly("epoch", &bflc::CommitteeLedger::epoch)
      .def_property_readonly("update_count",
                             &bflc::CommitteeLedger::update_count)
      .def_property_readonly("score_count",
                             &bflc::CommitteeLedger::score_count)
      .def_property_readonly("global_loss",
                             &bflc::CommitteeLedger::global_loss)
      .def("roles", &bflc::CommitteeLedger::roles)
      .def("registration_order", &bflc::CommitteeLedger::registration_order)
      .def("committee", &bflc::CommitteeLedger::committee)
      .def("trainers", &bflc::CommitteeLedger::trainers)
      .def("local_scores", &bflc::CommitteeLedger::local_scores)
      .def("update_blob", &bflc::CommitteeLedger::update_blob)
      .def("update_meta", &bflc::CommitteeLedger::update_meta)
      .def("has_pending", &bflc::CommitteeLedger::has_pending)
      .def_property_readonly("finished", &bflc::CommitteeLedger::finished)
      .def("snapshot", &bflc::CommitteeLedger::snapshot)
      .def("restore", &bflc::CommitteeLedger::restore);
}
