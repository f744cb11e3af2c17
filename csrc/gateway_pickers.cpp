// Native endpoint-picker core for the Gateway API Inference Extension
// integration.
//
// Parity target: reference src/gateway_inference_extension/
// {prefix_aware_picker.go (chunkSize=128, RWMutex hash trie),
//  kv_aware_picker.go (score table + round-robin fallback),
//  roundrobin_picker.go}.  The reference ships these as compiled Go
// plugins; here the same algorithms are compiled C++ exposed to the
// Python ext-proc/HTTP framing via pybind11 — the per-request pick path
// (hashing, trie walk, set intersection) runs without the GIL.
//
// Semantics match production_stack_amd/gateway/extproc.py::Picker exactly
// (same fallback order, same rr advance-before-use, same sorted-first
// tie-break, same seed-the-trie-after-fallback behaviour) so the Python
// implementation doubles as the differential test oracle.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <atomic>
#include <cstdint>
#include <memory>
#include <mutex>
#include <shared_mutex>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace py = pybind11;

namespace {

// FNV-1a 64-bit over one chunk of the prompt.  (The reference trie uses
// xxhash64; any 64-bit avalanche hash gives the same routing behaviour —
// keys never leave the process.)
static inline uint64_t chunk_hash(const char* p, size_t n) {
  uint64_t h = 1469598103934665603ull;
  for (size_t i = 0; i < n; ++i) {
    h ^= static_cast<uint8_t>(p[i]);
    h *= 1099511628211ull;
  }
  return h;
}

struct TrieNode {
  std::unordered_map<uint64_t, std::unique_ptr<TrieNode>> children;
  std::unordered_set<std::string> endpoints;
};

class PrefixTrie {
 public:
  explicit PrefixTrie(size_t chunk_size) : chunk_(chunk_size) {}

  void insert(const std::string& text, const std::string& endpoint) {
    std::unique_lock<std::shared_mutex> lk(mu_);
    TrieNode* node = &root_;
    for (size_t off = 0; off < text.size(); off += chunk_) {
      size_t n = std::min(chunk_, text.size() - off);
      uint64_t h = chunk_hash(text.data() + off, n);
      auto& child = node->children[h];
      if (!child) child = std::make_unique<TrieNode>();
      node = child.get();
      node->endpoints.insert(endpoint);
    }
  }

  // Longest prefix (in characters, whole chunks) for which at least one
  // of `live` endpoints is present; returns (matched_chars, candidates).
  std::pair<size_t, std::vector<std::string>> longest_prefix_match(
      const std::string& text, const std::vector<std::string>& live) const {
    std::unordered_set<std::string> live_set(live.begin(), live.end());
    std::shared_lock<std::shared_mutex> lk(mu_);
    const TrieNode* node = &root_;
    size_t matched = 0;
    std::vector<std::string> cands;
    for (size_t off = 0; off < text.size(); off += chunk_) {
      size_t n = std::min(chunk_, text.size() - off);
      uint64_t h = chunk_hash(text.data() + off, n);
      auto it = node->children.find(h);
      if (it == node->children.end()) break;
      node = it->second.get();
      std::vector<std::string> here;
      for (const auto& ep : node->endpoints)
        if (live_set.count(ep)) here.push_back(ep);
      if (here.empty()) break;
      matched = off + n;
      cands = std::move(here);
    }
    std::sort(cands.begin(), cands.end());
    return {matched, cands};
  }

  void remove_endpoint(const std::string& endpoint) {
    std::unique_lock<std::shared_mutex> lk(mu_);
    remove_rec(&root_, endpoint);
  }

 private:
  static void remove_rec(TrieNode* node, const std::string& endpoint) {
    node->endpoints.erase(endpoint);
    for (auto it = node->children.begin(); it != node->children.end();) {
      remove_rec(it->second.get(), endpoint);
      if (it->second->endpoints.empty() && it->second->children.empty())
        it = node->children.erase(it);
      else
        ++it;
    }
  }

  mutable std::shared_mutex mu_;
  TrieNode root_;
  size_t chunk_;
};

class NativePicker {
 public:
  NativePicker(size_t chunk_size, size_t min_match)
      : trie_(chunk_size), min_match_(min_match) {}

  std::string pick_roundrobin(const std::vector<std::string>& pods) {
    if (pods.empty()) return "";
    uint64_t i = ++rr_;
    return pods[i % pods.size()];
  }

  // prefixaware: longest match >= min_match wins (sorted-first
  // tie-break); otherwise round-robin and seed the trie with the pod
  // actually chosen, exactly like extproc.Picker._prefix_pick.
  std::string pick_prefixaware(const std::string& prompt,
                               const std::vector<std::string>& pods) {
    if (pods.empty()) return "";
    auto [matched, cands] = trie_.longest_prefix_match(prompt, pods);
    std::string pick;
    if (matched >= min_match_ && !cands.empty())
      pick = cands.front();
    else
      pick = pick_roundrobin(pods);
    trie_.insert(prompt, pick);
    return pick;
  }

  // kvaware: caller supplies {endpoint: matched_tokens} scores from the
  // KV controller lookup; best positive score among live pods wins,
  // else round-robin (reference kv_aware_picker.go fallback).
  std::string pick_kvaware(
      const std::unordered_map<std::string, int64_t>& scores,
      const std::vector<std::string>& pods) {
    if (pods.empty()) return "";
    std::string best;
    int64_t best_score = 0;
    for (const auto& p : pods) {
      auto it = scores.find(p);
      if (it != scores.end() && it->second > best_score) {
        best_score = it->second;
        best = p;
      }
    }
    if (!best.empty()) return best;
    return pick_roundrobin(pods);
  }

  void remove_endpoint(const std::string& ep) { trie_.remove_endpoint(ep); }

 private:
  PrefixTrie trie_;
  std::atomic<uint64_t> rr_{0};
  size_t min_match_;
};

}  // namespace

PYBIND11_MODULE(_gwpick, m) {
  m.doc() = "native gateway endpoint pickers (prefix trie / kv / rr)";
  py::class_<PrefixTrie>(m, "PrefixTrie")
      .def(py::init<size_t>(), py::arg("chunk_size") = 128)
      .def("insert", &PrefixTrie::insert,
           py::call_guard<py::gil_scoped_release>())
      .def("longest_prefix_match", &PrefixTrie::longest_prefix_match,
           py::call_guard<py::gil_scoped_release>())
      .def("remove_endpoint", &PrefixTrie::remove_endpoint,
           py::call_guard<py::gil_scoped_release>());
  py::class_<NativePicker>(m, "NativePicker")
      .def(py::init<size_t, size_t>(), py::arg("chunk_size") = 128,
           py::arg("min_match") = 128)
      .def("pick_roundrobin", &NativePicker::pick_roundrobin,
           py::call_guard<py::gil_scoped_release>())
      .def("pick_prefixaware", &NativePicker::pick_prefixaware,
           py::call_guard<py::gil_scoped_release>())
      .def("pick_kvaware", &NativePicker::pick_kvaware,
           py::call_guard<py::gil_scoped_release>())
      .def("remove_endpoint", &NativePicker::remove_endpoint,
           py::call_guard<py::gil_scoped_release>());
}
