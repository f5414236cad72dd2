// Native call-path featurizer (CPU).
//
// The featurization DFS over span trees is the host-side hot loop for large
// apps (4096 endpoints x many windows — SURVEY.md 3.1).  Instead of building
// a path key per node (the reference stringifies the whole prefix per node,
// featurize.py:14), we walk a TRIE of call paths in lockstep with the span
// tree: one hash lookup per span on "component\0operation", O(spans) total.
//
// Exposed as deeprest_amd._C.featurize_walk(windows, seed_paths, grow):
//   windows:    list of {'traces': [span-tree...]} dicts (contract format)
//   seed_paths: list of ((component, operation), ...) tuples — the existing
//               feature space, trie-ified first so indices line up
//   grow:       discover new paths (fit_transform) or count only (transform)
// returns (new_paths, traffic (T,P) int64 tensor, inv_names,
//          inv (len(inv_names), T) int64 tensor) — new_paths only lists
// paths beyond the seed, in discovery order.
#include <torch/extension.h>

#include <functional>
#include <memory>
#include <string>
#include <unordered_map>
#include <vector>

namespace {

struct TrieNode {
  std::unordered_map<std::string, std::unique_ptr<TrieNode>> children;
  int index = -1;
};

std::string edge_key(const std::string& comp, const std::string& op) {
  std::string k;
  k.reserve(comp.size() + op.size() + 1);
  k += comp;
  k += '\0';
  k += op;
  return k;
}

py::object featurize_walk(py::list windows, py::list seed_paths, bool grow) {
  TrieNode root;
  int next_index = 0;
  // path index -> (component, operation) chain, for reporting new paths
  std::vector<std::vector<std::pair<std::string, std::string>>> all_paths;

  // ---- seed the trie from the existing feature space ----
  for (auto path_obj : seed_paths) {
    py::sequence path = path_obj.cast<py::sequence>();
    TrieNode* node = &root;
    std::vector<std::pair<std::string, std::string>> chain;
    for (auto pair_obj : path) {
      py::sequence pr = pair_obj.cast<py::sequence>();
      std::string comp = pr[0].cast<std::string>();
      std::string op = pr[1].cast<std::string>();
      chain.emplace_back(comp, op);
      std::string key = edge_key(comp, op);
      auto it = node->children.find(key);
      if (it == node->children.end())
        it = node->children.emplace(key, std::make_unique<TrieNode>()).first;
      node = it->second.get();
    }
    if (node->index < 0) {
      node->index = next_index++;
      all_paths.push_back(std::move(chain));
    }
  }
  const int n_seed = next_index;

  const int64_t T = (int64_t)windows.size();
  py::str k_traces("traces"), k_children("children");
  py::str k_component("component"), k_operation("operation");

  // ---- pass 1 (grow mode): discover new paths in window/trace order ----
  struct Frame {
    PyObject* span;
    TrieNode* node;  // trie node of the span's PARENT path
  };

  std::vector<Frame> stack;
  if (grow) {
    for (int64_t t = 0; t < T; ++t) {
      py::dict w = windows[t].cast<py::dict>();
      py::list traces = w[k_traces].cast<py::list>();
      for (auto tr : traces) {
        stack.push_back({tr.ptr(), &root});
        while (!stack.empty()) {
          Frame f = stack.back();
          stack.pop_back();
          py::dict span = py::reinterpret_borrow<py::dict>(f.span);
          std::string comp = span[k_component].cast<std::string>();
          std::string op = span[k_operation].cast<std::string>();
          std::string key = edge_key(comp, op);
          auto it = f.node->children.find(key);
          if (it == f.node->children.end())
            it = f.node->children.emplace(key, std::make_unique<TrieNode>()).first;
          TrieNode* node = it->second.get();
          if (node->index < 0) {
            node->index = next_index++;
            // reconstructing the chain lazily is awkward; store parent chain
            // via the path list of the parent + this pair.  Simplest: rebuild
            // from scratch is O(depth); track with a side walk below.
            all_paths.emplace_back();  // placeholder, filled after
          }
          py::list children = span[k_children].cast<py::list>();
          // push reversed for left-to-right discovery order
          for (Py_ssize_t i = py::len(children) - 1; i >= 0; --i)
            stack.push_back({PyList_GET_ITEM(children.ptr(), i), node});
        }
      }
    }
    // fill in the textual form of newly discovered paths by walking the trie
    // (iterative — deep path chains must not overflow the C++ stack)
    {
      std::vector<std::pair<std::string, std::string>> chain;
      struct WFrame {
        TrieNode* node;
        std::unordered_map<std::string, std::unique_ptr<TrieNode>>::iterator it;
      };
      std::vector<WFrame> wstack;
      wstack.push_back({&root, root.children.begin()});
      while (!wstack.empty()) {
        WFrame& top = wstack.back();
        if (top.it == top.node->children.end()) {
          wstack.pop_back();
          if (!chain.empty()) chain.pop_back();
          continue;
        }
        const std::string& key = top.it->first;
        TrieNode* child = top.it->second.get();
        ++top.it;
        size_t z = key.find('\0');
        chain.emplace_back(key.substr(0, z), key.substr(z + 1));
        if (child->index >= n_seed) all_paths[child->index] = chain;
        wstack.push_back({child, child->children.begin()});
      }
    }
  }

  const int64_t P = next_index;

  // ---- pass 2: count paths + component invocations per window ----
  auto traffic = torch::zeros({T, P}, torch::kInt64);
  auto* traffic_ptr = traffic.data_ptr<int64_t>();
  std::unordered_map<std::string, int> comp_index;
  std::vector<std::string> comp_names;
  std::vector<std::vector<int64_t>> inv_cols;  // per component, length T
  auto comp_slot = [&](const std::string& c) -> int64_t* {
    auto it = comp_index.find(c);
    if (it == comp_index.end()) {
      it = comp_index.emplace(c, (int)comp_names.size()).first;
      comp_names.push_back(c);
      inv_cols.emplace_back(T, 0);
    }
    return inv_cols[it->second].data();
  };
  int64_t* general = comp_slot("general");

  // sentinel for spans whose path left the trie: since the path set is
  // prefix-closed, no descendant of an unknown path can be a known path
  TrieNode dead;
  for (int64_t t = 0; t < T; ++t) {
    py::dict w = windows[t].cast<py::dict>();
    py::list traces = w[k_traces].cast<py::list>();
    int64_t* row = traffic_ptr + t * P;
    for (auto tr : traces) {
      inv_cols[comp_index.at("general")][t] += 1;
      stack.push_back({tr.ptr(), &root});
      while (!stack.empty()) {
        Frame f = stack.back();
        stack.pop_back();
        py::dict span = py::reinterpret_borrow<py::dict>(f.span);
        std::string comp = span[k_component].cast<std::string>();
        std::string op = span[k_operation].cast<std::string>();
        comp_slot(comp)[t] += 1;
        TrieNode* node = &dead;
        auto it = f.node->children.find(edge_key(comp, op));
        if (it != f.node->children.end()) {
          node = it->second.get();
          if (node->index >= 0) row[node->index] += 1;
        }
        py::list children = span[k_children].cast<py::list>();
        for (Py_ssize_t i = py::len(children) - 1; i >= 0; --i)
          stack.push_back({PyList_GET_ITEM(children.ptr(), i), node});
      }
    }
  }
  (void)general;

  // ---- assemble outputs ----
  py::list new_paths;
  for (int i = n_seed; i < next_index; ++i) {
    py::tuple path(all_paths[i].size());
    for (size_t d = 0; d < all_paths[i].size(); ++d)
      path[d] = py::make_tuple(all_paths[i][d].first, all_paths[i][d].second);
    new_paths.append(path);
  }
  py::list inv_names;
  auto inv = torch::zeros({(int64_t)comp_names.size(), T}, torch::kInt64);
  auto* inv_ptr = inv.data_ptr<int64_t>();
  for (size_t c = 0; c < comp_names.size(); ++c) {
    inv_names.append(py::str(comp_names[c]));
    std::copy(inv_cols[c].begin(), inv_cols[c].end(), inv_ptr + c * T);
  }
  return py::make_tuple(new_paths, traffic, inv_names, inv);
}

}  // namespace

void register_featurize(py::module_& m) {
  m.def("featurize_walk", &featurize_walk, py::arg("windows"),
        py::arg("seed_paths"), py::arg("grow"));
}
