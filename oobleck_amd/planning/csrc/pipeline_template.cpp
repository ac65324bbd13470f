// pipeline_template.cpp — pybind-identical rebuild of the reference's C++
// planner (SURVEY.md §2 "Planner": kept "as-is ... or re-exposed with
// identical pybind API").  The reference module
// (/root/reference/oobleck/csrc/planning/pipeline_template.cpp:82-161,
// pipeline_template.h:57-84, execution_result.h:114-204, bind.cpp:11-69)
// depends on cppcoro + oneTBB + nlohmann/json, all of which are ABSENT from
// this environment (empty un-vendored submodules, no system packages, no
// network) — so the module cannot compile verbatim.  This file restates the
// SAME algorithm behind the SAME pybind surface (pipeline_template.pyi is
// the contract), replacing:
//   cppcoro task pool   -> std::thread fan-out over the top-level
//                          (num_nodes, num_stages) jobs, shared memo table
//                          under a shared_mutex (duplicate computation of a
//                          key is benign-idempotent, like the reference's
//                          TBB concurrent map),
//   nlohmann::json      -> the embedded interpreter's own `json` module
//                          (get_profile_results is called from Python),
//   oneTBB memo map     -> std::unordered_map + std::shared_mutex.
//
// Algorithm semantics restated 1:1 (Oobleck paper §4.1.2):
//   * StageExecutionResult aggregation: fwd/bwd divided by the stage's GPU
//     count, + allreduce_in_node[num_gpus] when sharded, mem = 6*params +
//     activations (execution_result.h:66-96).
//   * DC combine: t1 = sum of per-stage fwd+bwd; kstar = the slower side's
//     bottleneck stage (right's index shifted by left's stage count);
//     t2 = (2*num_stages + kstar + 1) * kstar_latency; t3 = the fwd+bwd sum
//     from kstar to the end (both sides when kstar is on the left)
//     (execution_result.h:114-204).
//   * Divide: split the layer range at every k; nodes>1 split nodes,
//     nodes==1 split the node's GPUs — only into EXACT HALVES, the
//     reference's quirk (pipeline_template.cpp:240-244), kept verbatim;
//     1-stage templates require power-of-two GPU counts (:196-200).
//   * Infeasibility rules and memoized nullptr results kept identical.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <atomic>
#include <cmath>
#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <shared_mutex>
#include <string>
#include <thread>
#include <tuple>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

namespace obplan {

struct LayerExecutionResult {
  int layer_index;
  double forward;
  double backward;
  std::map<int, double> allreduce_in_node;
  std::map<int, double> allreduce_across_nodes;
  std::tuple<int, int> mem_required;
};

class LayerExecutionResults {
 public:
  explicit LayerExecutionResults(std::vector<LayerExecutionResult>&& data)
      : data_(std::move(data)) {}
  const std::vector<LayerExecutionResult>& get() const { return data_; }
  const LayerExecutionResult& at(int i) const { return data_.at(i); }
  int size() const { return (int)data_.size(); }

 private:
  std::vector<LayerExecutionResult> data_;
};

struct StageExecutionResult {
  // aggregation semantics of execution_result.h:66-96
  StageExecutionResult(const std::shared_ptr<LayerExecutionResults>& layers,
                       const std::tuple<int, int>& layer_range, int num_gpus)
      : num_gpus(num_gpus) {
    const int lo = std::get<0>(layer_range), hi = std::get<1>(layer_range);
    for (int i = lo; i < hi; ++i) {
      const auto& l = layers->at(i);
      layer_indices.push_back(l.layer_index);
      forward += l.forward / num_gpus;
      backward += l.backward / num_gpus;
      if (num_gpus > 1) {
        // intra-stage sharding adds the in-node allreduce both ways
        forward += l.allreduce_in_node.at(num_gpus);
        backward += l.allreduce_in_node.at(num_gpus);
      }
      for (const auto& [n, t] : l.allreduce_across_nodes)
        allreduce_across_nodes[n] += t;
      mem_required += std::get<0>(l.mem_required) * 6;
      mem_required += std::get<1>(l.mem_required);
    }
  }
  int num_layers() const { return (int)layer_indices.size(); }
  double fb() const { return forward + backward; }

  int num_gpus;
  std::vector<int> layer_indices;
  double forward = 0.0;
  double backward = 0.0;
  std::map<int, double> allreduce_across_nodes;
  int mem_required = 0;
};

using StagePtr = std::shared_ptr<StageExecutionResult>;

// One memoized divide-and-conquer solution: the 1F1B iteration-latency
// model t1+t2+t3 with the bottleneck stage kstar (execution_result.h:
// 114-204; Oobleck paper eq. for pipeline iteration time).
struct DCResult {
  double t1, t2, t3;
  int kstar;  // index of the bottleneck stage within `stages`
  std::vector<StagePtr> stages;

  static std::shared_ptr<DCResult> single(const StagePtr& s) {
    auto r = std::make_shared<DCResult>();
    r->t1 = s->fb();
    r->t2 = 2 * s->fb();
    r->t3 = s->fb();
    r->kstar = 0;
    r->stages = {s};
    return r;
  }

  double kstar_latency() const { return stages[kstar]->fb(); }
  double total() const { return t1 + t2 + t3; }

  static std::shared_ptr<DCResult> combine(
      const std::shared_ptr<DCResult>& a, const std::shared_ptr<DCResult>& b) {
    auto r = std::make_shared<DCResult>();
    r->stages = a->stages;
    r->stages.insert(r->stages.end(), b->stages.begin(), b->stages.end());
    r->kstar = a->kstar_latency() > b->kstar_latency()
                   ? a->kstar
                   : b->kstar + (int)a->stages.size();
    r->t1 = a->t1 + b->t1;
    const int nk =
        2 * (int)(a->stages.size() + b->stages.size()) + r->kstar + 1;
    double tail = 0.0;
    if (r->kstar == a->kstar) {
      r->t2 = nk * a->kstar_latency();
      for (size_t i = a->kstar; i < a->stages.size(); ++i)
        tail += a->stages[i]->fb();
      for (const auto& s : b->stages) tail += s->fb();
    } else {
      r->t2 = nk * b->kstar_latency();
      for (size_t i = b->kstar; i < b->stages.size(); ++i)
        tail += b->stages[i]->fb();
    }
    r->t3 = tail;
    return r;
  }
};

class PipelineTemplate {
 public:
  PipelineTemplate(const std::vector<StagePtr>& stages, double iteration_time,
                   int num_layers, int num_nodes, int num_gpus_per_node)
      : stages_(stages),
        iteration_time_(iteration_time),
        num_nodes_(num_nodes),
        num_gpus_per_node_(num_gpus_per_node) {
    int gpus = 0, nlayers = 0;
    for (const auto& s : stages_) {
      gpus += s->num_gpus;
      nlayers += s->num_layers();
    }
    if (gpus != num_nodes * num_gpus_per_node)
      throw py::value_error("stages use " + std::to_string(gpus) +
                            " GPUs, template declares " +
                            std::to_string(num_nodes * num_gpus_per_node));
    if (nlayers != num_layers)
      throw py::value_error("stages cover " + std::to_string(nlayers) +
                            " layers, template declares " +
                            std::to_string(num_layers));
  }

  double iteration_time() const { return iteration_time_; }
  const std::vector<StagePtr>& get_stages() const { return stages_; }
  int num_nodes() const { return num_nodes_; }
  int num_gpus_per_node() const { return num_gpus_per_node_; }

  // layer index -> per-fsdp-slot rank list (pipeline_template.h:57-84):
  // each stage consumes its num_gpus ranks off the front of `ranks`; a
  // stage with fewer GPUs than num_gpus_per_node repeats each rank
  // (num_gpus_per_node / stage_gpus) times so every layer's list has
  // num_gpus_per_node entries.
  std::map<int, std::vector<int>> get_rank_grid(std::vector<int> ranks) const {
    std::map<int, std::vector<int>> grid;
    size_t cursor = 0;
    for (const auto& s : stages_) {
      if (cursor + s->num_gpus > ranks.size())
        throw py::value_error("get_rank_grid: not enough ranks");
      std::vector<int> per_layer;
      per_layer.reserve(num_gpus_per_node_);
      const int repeat = num_gpus_per_node_ / s->num_gpus;
      for (int g = 0; g < s->num_gpus; ++g)
        for (int r = 0; r < repeat; ++r)
          per_layer.push_back(ranks[cursor + g]);
      cursor += s->num_gpus;
      for (int lid : s->layer_indices) grid[lid] = per_layer;
    }
    if (cursor != ranks.size())
      throw py::value_error("get_rank_grid: leftover ranks");
    return grid;
  }

 private:
  std::vector<StagePtr> stages_;
  double iteration_time_;
  int num_nodes_;
  int num_gpus_per_node_;
};

// NOTE: like the reference's dc_cache_ (execution_result.h:213), the memo
// table is keyed by (num_stages, layer range, num_nodes, num_gpus) WITHOUT
// the profile's identity — a generator instance serves ONE profile.
class PipelineTemplateGenerator {
 public:
  std::vector<PipelineTemplate> create_pipeline_templates(
      std::shared_ptr<LayerExecutionResults> layers,
      const std::tuple<int, int>& node_range, int num_gpus_per_node) {
    const int lo = std::get<0>(node_range), hi = std::get<1>(node_range);
    std::vector<PipelineTemplate> out;
    // one job per node count; each job scans its feasible stage counts.
    // jobs share the memo table (benign duplicate computation, like the
    // reference's TBB map).  GIL released while the pool runs.
    std::vector<std::shared_ptr<DCResult>> best(hi - lo + 1, nullptr);
    {
      py::gil_scoped_release release;
      std::vector<std::thread> pool;
      for (int n = lo; n <= hi; ++n) {
        pool.emplace_back([this, &layers, &best, n, lo, num_gpus_per_node] {
          std::shared_ptr<DCResult> opt = nullptr;
          for (int stages = n; stages <= layers->size(); ++stages) {
            auto r = solve(layers, 0, layers->size(), stages, n,
                           num_gpus_per_node);
            if (r && (!opt || r->total() < opt->total())) opt = r;
          }
          best[n - lo] = opt;
        });
      }
      for (auto& t : pool) t.join();
    }
    for (int n = lo; n <= hi; ++n) {
      const auto& opt = best[n - lo];
      if (!opt) continue;  // no feasible template for this node count
      out.emplace_back(opt->stages, opt->total(), layers->size(), n,
                       num_gpus_per_node);
    }
    return out;
  }

  unsigned long cache_hits() const { return hits_.load(); }
  unsigned long cache_misses() const { return misses_.load(); }

 private:
  // (num_stages, start, end, num_nodes, num_gpus_per_node)
  using Key = std::tuple<int, int, int, int, int>;
  struct KeyHash {
    size_t operator()(const Key& k) const {
      uint64_t h = 1469598103934665603ull;
      auto mix = [&h](uint64_t v) {
        h ^= v + 0x9e3779b97f4a7c15ull + (h << 6) + (h >> 2);
      };
      mix(std::get<0>(k));
      mix(std::get<1>(k));
      mix(std::get<2>(k));
      mix(std::get<3>(k));
      mix(std::get<4>(k));
      return (size_t)h;
    }
  };

  std::unordered_map<Key, std::shared_ptr<DCResult>, KeyHash> memo_;
  std::shared_mutex memo_mu_;
  std::atomic<unsigned long> hits_{0}, misses_{0};

  bool lookup(const Key& k, std::shared_ptr<DCResult>& out) {
    std::shared_lock lk(memo_mu_);
    auto it = memo_.find(k);
    if (it == memo_.end()) return false;
    out = it->second;
    return true;
  }

  void store(const Key& k, const std::shared_ptr<DCResult>& v) {
    std::unique_lock lk(memo_mu_);
    memo_.emplace(k, v);
  }

  std::shared_ptr<DCResult> solve(
      const std::shared_ptr<LayerExecutionResults>& layers, int start,
      int end, int num_stages, int num_nodes, int num_gpus_per_node) {
    const Key key{num_stages, start, end, num_nodes, num_gpus_per_node};
    std::shared_ptr<DCResult> cached;
    if (lookup(key, cached)) {
      hits_.fetch_add(1, std::memory_order_relaxed);
      return cached;
    }
    misses_.fetch_add(1, std::memory_order_relaxed);

    // infeasibility rules (pipeline_template.cpp:190-211)
    bool infeasible = false;
    if (num_stages > end - start) infeasible = true;
    if (num_nodes == 1) {
      if (num_gpus_per_node < num_stages) infeasible = true;
      const double lg = std::log2((double)num_gpus_per_node);
      if (num_stages == 1 && lg != std::trunc(lg)) infeasible = true;
    } else if (num_nodes > num_stages) {
      infeasible = true;  // >=2 nodes cannot share one stage
    }
    if (infeasible) {
      store(key, nullptr);
      return nullptr;
    }

    if (num_stages == 1) {
      auto stage = std::make_shared<StageExecutionResult>(
          layers, std::make_tuple(start, end), num_gpus_per_node);
      auto r = DCResult::single(stage);
      store(key, r);
      return r;
    }

    std::shared_ptr<DCResult> result = nullptr;
    auto consider = [&](const std::shared_ptr<DCResult>& l,
                        const std::shared_ptr<DCResult>& r) {
      if (!l || !r) return;
      auto c = DCResult::combine(l, r);
      if (!result || c->total() < result->total()) result = c;
    };
    for (int k = start + 1; k < end; ++k) {
      if (num_nodes == 1) {
        // split the node's GPUs — EXACT halves only (reference quirk,
        // pipeline_template.cpp:240-244)
        for (int gl = 1; gl < num_gpus_per_node; ++gl) {
          if (gl != num_gpus_per_node - gl) continue;
          for (int sl = 1; sl < num_stages; ++sl)
            consider(solve(layers, start, k, sl, 1, gl),
                     solve(layers, k, end, num_stages - sl, 1,
                           num_gpus_per_node - gl));
        }
      } else {
        for (int nl = 1; nl < num_nodes; ++nl)
          for (int sl = 1; sl < num_stages; ++sl)
            consider(solve(layers, start, k, sl, nl, num_gpus_per_node),
                     solve(layers, k, end, num_stages - sl, num_nodes - nl,
                           num_gpus_per_node));
      }
    }
    store(key, result);
    return result;
  }
};

// get_profile_results (pipeline_template.cpp:29-80): loads the profiler's
// JSON cache from /tmp/oobleck/profiles/<model>-<tag>/.  JSON parsing via
// the embedded interpreter's own json module (nlohmann is absent).
std::shared_ptr<LayerExecutionResults> get_profile_results(
    const std::string& model_name, const std::string& model_tag,
    int microbatch_size) {
  py::module_ json = py::module_::import("json");
  py::module_ builtins = py::module_::import("builtins");
  const std::string base =
      "/tmp/oobleck/profiles/" + model_name + "-" + model_tag + "/";
  auto load = [&](const std::string& path) {
    py::object f = builtins.attr("open")(path);
    py::object data = json.attr("load")(f);
    f.attr("close")();
    return data;
  };
  py::list mb = load(base + "mb" + std::to_string(microbatch_size) + ".json");
  py::list ar_in = load(base + "allreduce_in_node.json");
  py::list ar_across = load(base + "allreduce_across_nodes.json");

  std::vector<LayerExecutionResult> out;
  const int n = (int)py::len(mb);
  for (int i = 0; i < n; ++i) {
    LayerExecutionResult l;
    l.layer_index = i;
    py::dict row = mb[i];
    l.forward = row["forward"].cast<double>();
    l.backward = row["backward"].cast<double>();
    for (auto item : ar_in[i].cast<py::dict>())
      l.allreduce_in_node[std::stoi(item.first.cast<std::string>())] =
          item.second.cast<double>();
    for (auto item : ar_across[i].cast<py::dict>())
      l.allreduce_across_nodes[std::stoi(item.first.cast<std::string>())] =
          item.second.cast<double>();
    l.mem_required = row["mem_required"].cast<std::tuple<int, int>>();
    out.push_back(std::move(l));
  }
  return std::make_shared<LayerExecutionResults>(std::move(out));
}

}  // namespace obplan

using namespace obplan;

PYBIND11_MODULE(pipeline_template, m) {
  m.doc() =
      "MI355X rebuild of Oobleck's planner (pybind API per "
      "pipeline_template.pyi)";

  py::class_<LayerExecutionResult>(m, "LayerExecutionResult")
      .def(py::init([](int layer_index, double forward, double backward,
                       const std::map<int, double>& allreduce_in_node,
                       const std::map<int, double>& allreduce_across_nodes,
                       const std::tuple<int, int>& mem_required) {
             return LayerExecutionResult{layer_index,  forward,
                                         backward,     allreduce_in_node,
                                         allreduce_across_nodes, mem_required};
           }),
           py::arg("layer_index"), py::arg("forward"), py::arg("backward"),
           py::arg("allreduce_in_node"), py::arg("allreduce_across_nodes"),
           py::arg("mem_required"))
      .def_readonly("_index", &LayerExecutionResult::layer_index)
      .def_readonly("_forward", &LayerExecutionResult::forward)
      .def_readonly("_backward", &LayerExecutionResult::backward)
      .def_readonly("_allreduce_in_node",
                    &LayerExecutionResult::allreduce_in_node)
      .def_readonly("_allreduce_across_nodes",
                    &LayerExecutionResult::allreduce_across_nodes)
      .def_readonly("_mem_required", &LayerExecutionResult::mem_required);

  py::class_<LayerExecutionResults, std::shared_ptr<LayerExecutionResults>>(
      m, "LayerExecutionResults")
      .def(py::init<std::vector<LayerExecutionResult>&&>())
      .def("get", &LayerExecutionResults::get)
      .def("at", &LayerExecutionResults::at, py::arg("index"))
      .def_property_readonly("size", &LayerExecutionResults::size);

  py::class_<StageExecutionResult, std::shared_ptr<StageExecutionResult>>(
      m, "StageExecutionResult")
      .def(py::init<const std::shared_ptr<LayerExecutionResults>&,
                    const std::tuple<int, int>&, int>())
      .def_readonly("_num_gpus", &StageExecutionResult::num_gpus)
      .def_readonly("_layer_indices", &StageExecutionResult::layer_indices)
      .def_readonly("_forward", &StageExecutionResult::forward)
      .def_readonly("_backward", &StageExecutionResult::backward)
      .def_property_readonly("_num_layers", &StageExecutionResult::num_layers)
      .def_readonly("_mem_required", &StageExecutionResult::mem_required);

  py::class_<PipelineTemplate>(m, "PipelineTemplate")
      .def(py::init<const std::vector<StagePtr>&, double, int, int, int>())
      .def("get_stages", &PipelineTemplate::get_stages)
      .def("get_rank_grid", &PipelineTemplate::get_rank_grid, py::arg("ranks"))
      .def_property_readonly("_iteration_time",
                             &PipelineTemplate::iteration_time)
      .def_property_readonly("_num_nodes", &PipelineTemplate::num_nodes)
      .def_property_readonly("_num_gpus_per_node",
                             &PipelineTemplate::num_gpus_per_node)
      .def("__repr__", [](const PipelineTemplate& pt) {
        return "<oobleck.PipelineTemplate." +
               std::to_string(pt.num_nodes()) + "nodes>";
      });

  py::class_<PipelineTemplateGenerator>(m, "PipelineTemplateGenerator")
      .def(py::init<>())
      .def("create_pipeline_templates",
           &PipelineTemplateGenerator::create_pipeline_templates);

  m.def("get_profile_results", &get_profile_results, py::arg("model_name"),
        py::arg("model_tag"), py::arg("microbatch_size"));
}
