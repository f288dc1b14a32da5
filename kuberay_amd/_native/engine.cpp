// engine — native C++ object-store backend for the kuberay-amd control plane.
//
// The reconcile path of the operator is bound by its object cache: at the
// 500-cluster soak the cache holds ~2500 objects (2000 pods),
// and a Python dict-tree store pays deep copies + GC pressure + RSS for all
// of them. This backend keeps every object as ONE compact JSON blob in C++
// heap with label / kind / owner-uid indexes beside it, plus precomputed
// "pod views" (the 9 fields the RayCluster reconciler actually reads per
// pod per reconcile) so the hot loop never materializes a pod at all.
//
// Semantics live in python (kuberay_amd/kube/store.py InMemoryApiServer);
// this is pure storage + indexing. Thread-safe via an internal mutex —
// callers do NOT need the python-side lock for reads.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cstdint>
#include <mutex>
#include <optional>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace py = pybind11;

namespace {

struct PodViewData {
  std::string name, ns, phase, deletion_ts, pod_ip, restart_policy, creation_ts;
  bool ready = false;
  bool terminated = false;
};

struct Entry {
  std::string kind, ns, name;
  std::string blob;  // compact JSON of the whole object
  std::string rv;
  std::vector<std::pair<std::string, std::string>> labels;
  std::vector<std::string> owner_uids;
  bool has_view = false;
  PodViewData view;
};

inline std::string make_key(const std::string& kind, const std::string& ns,
                            const std::string& name) {
  std::string k;
  k.reserve(kind.size() + ns.size() + name.size() + 2);
  k += kind; k += '\x1f'; k += ns; k += '\x1f'; k += name;
  return k;
}

inline std::string label_key(const std::string& kind, const std::string& k,
                             const std::string& v) {
  std::string out;
  out.reserve(kind.size() + k.size() + v.size() + 2);
  out += kind; out += '\x1f'; out += k; out += '\x1f'; out += v;
  return out;
}

class NativeStore {
 public:
  void put(const std::string& kind, const std::string& ns,
           const std::string& name, const std::string& blob,
           const std::string& rv,
           const std::vector<std::pair<std::string, std::string>>& labels,
           const std::vector<std::string>& owner_uids,
           const std::optional<PodViewData>& view) {
    std::lock_guard<std::mutex> g(mu_);
    const std::string key = make_key(kind, ns, name);
    auto it = objects_.find(key);
    if (it != objects_.end()) {
      index_remove(key, it->second);
    }
    Entry e;
    e.kind = kind; e.ns = ns; e.name = name;
    e.blob = blob; e.rv = rv;
    e.labels = labels; e.owner_uids = owner_uids;
    if (view) { e.has_view = true; e.view = *view; }
    index_add(key, e);
    objects_[key] = std::move(e);
  }

  std::optional<py::bytes> fetch(const std::string& kind, const std::string& ns,
                                 const std::string& name) const {
    std::lock_guard<std::mutex> g(mu_);
    auto it = objects_.find(make_key(kind, ns, name));
    if (it == objects_.end()) return std::nullopt;
    return py::bytes(it->second.blob);
  }

  std::optional<std::string> rv(const std::string& kind, const std::string& ns,
                                const std::string& name) const {
    std::lock_guard<std::mutex> g(mu_);
    auto it = objects_.find(make_key(kind, ns, name));
    if (it == objects_.end()) return std::nullopt;
    return it->second.rv;
  }

  bool contains(const std::string& kind, const std::string& ns,
                const std::string& name) const {
    std::lock_guard<std::mutex> g(mu_);
    return objects_.count(make_key(kind, ns, name)) != 0;
  }

  std::optional<py::bytes> remove(const std::string& kind, const std::string& ns,
                                  const std::string& name) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = objects_.find(make_key(kind, ns, name));
    if (it == objects_.end()) return std::nullopt;
    py::bytes out(it->second.blob);
    index_remove(it->first, it->second);
    objects_.erase(it);
    return out;
  }

  std::vector<py::bytes> list_blobs(
      const std::string& kind, const std::optional<std::string>& ns,
      const std::vector<std::pair<std::string, std::string>>& selector) const {
    std::lock_guard<std::mutex> g(mu_);
    std::vector<const Entry*> entries = select(kind, ns, selector);
    std::sort(entries.begin(), entries.end(), [](const Entry* a, const Entry* b) {
      return std::tie(a->ns, a->name) < std::tie(b->ns, b->name);
    });
    std::vector<py::bytes> out;
    out.reserve(entries.size());
    for (const Entry* e : entries) out.emplace_back(e->blob);
    return out;
  }

  // Returns per-pod view tuples:
  // (name, ns, labels, phase, ready, deletion_ts, pod_ip, restart_policy,
  //  terminated, creation_ts)
  py::list list_views(
      const std::optional<std::string>& ns,
      const std::vector<std::pair<std::string, std::string>>& selector) const {
    std::vector<const Entry*> entries;
    {
      std::lock_guard<std::mutex> g(mu_);
      entries = select("Pod", ns, selector);
      std::sort(entries.begin(), entries.end(), [](const Entry* a, const Entry* b) {
        return std::tie(a->ns, a->name) < std::tie(b->ns, b->name);
      });
      // build python objects while still holding the lock (entries point
      // into the map; cheap enough at view granularity)
      py::list out;
      for (const Entry* e : entries) {
        if (!e->has_view) continue;
        py::dict labels;
        for (const auto& kv : e->labels) {
          labels[py::str(kv.first)] = py::str(kv.second);
        }
        out.append(py::make_tuple(
            e->view.name, e->view.ns, labels, e->view.phase, e->view.ready,
            e->view.deletion_ts, e->view.pod_ip, e->view.restart_policy,
            e->view.terminated, e->view.creation_ts));
      }
      return out;
    }
  }

  std::vector<py::tuple> dependents(const std::string& uid) const {
    std::lock_guard<std::mutex> g(mu_);
    std::vector<py::tuple> out;
    auto it = owner_index_.find(uid);
    if (it == owner_index_.end()) return out;
    for (const auto& key : it->second) {
      auto oit = objects_.find(key);
      if (oit != objects_.end()) {
        out.push_back(py::make_tuple(oit->second.kind, oit->second.ns,
                                     oit->second.name));
      }
    }
    return out;
  }

  void drop_owner(const std::string& uid) {
    std::lock_guard<std::mutex> g(mu_);
    owner_index_.erase(uid);
  }

  size_t count(const std::string& kind) const {
    std::lock_guard<std::mutex> g(mu_);
    auto it = kind_index_.find(kind);
    return it == kind_index_.end() ? 0 : it->second.size();
  }

  size_t total_bytes() const {
    std::lock_guard<std::mutex> g(mu_);
    size_t n = 0;
    for (const auto& kv : objects_) n += kv.second.blob.size();
    return n;
  }

 private:
  std::vector<const Entry*> select(
      const std::string& kind, const std::optional<std::string>& ns,
      const std::vector<std::pair<std::string, std::string>>& selector) const {
    std::vector<const Entry*> out;
    const std::unordered_set<std::string>* keys = nullptr;
    std::unordered_set<std::string> intersection;
    if (!selector.empty()) {
      // start from the smallest label bucket
      const std::unordered_set<std::string>* smallest = nullptr;
      for (const auto& kv : selector) {
        auto it = label_index_.find(label_key(kind, kv.first, kv.second));
        if (it == label_index_.end()) return out;  // empty bucket -> no match
        if (smallest == nullptr || it->second.size() < smallest->size()) {
          smallest = &it->second;
        }
      }
      keys = smallest;
    } else {
      auto it = kind_index_.find(kind);
      if (it == kind_index_.end()) return out;
      keys = &it->second;
    }
    for (const auto& key : *keys) {
      auto oit = objects_.find(key);
      if (oit == objects_.end()) continue;
      const Entry& e = oit->second;
      if (e.kind != kind) continue;
      if (ns && e.ns != *ns) continue;
      bool ok = true;
      for (const auto& kv : selector) {
        bool found = false;
        for (const auto& lv : e.labels) {
          if (lv.first == kv.first && lv.second == kv.second) { found = true; break; }
        }
        if (!found) { ok = false; break; }
      }
      if (ok) out.push_back(&e);
    }
    return out;
  }

  void index_add(const std::string& key, const Entry& e) {
    kind_index_[e.kind].insert(key);
    for (const auto& kv : e.labels) {
      label_index_[label_key(e.kind, kv.first, kv.second)].insert(key);
    }
    for (const auto& uid : e.owner_uids) owner_index_[uid].insert(key);
  }

  void index_remove(const std::string& key, const Entry& e) {
    auto kit = kind_index_.find(e.kind);
    if (kit != kind_index_.end()) kit->second.erase(key);
    for (const auto& kv : e.labels) {
      auto lit = label_index_.find(label_key(e.kind, kv.first, kv.second));
      if (lit != label_index_.end()) {
        lit->second.erase(key);
        if (lit->second.empty()) label_index_.erase(lit);
      }
    }
    for (const auto& uid : e.owner_uids) {
      auto oit = owner_index_.find(uid);
      if (oit != owner_index_.end()) oit->second.erase(key);
    }
  }

  mutable std::mutex mu_;
  std::unordered_map<std::string, Entry> objects_;
  std::unordered_map<std::string, std::unordered_set<std::string>> kind_index_;
  std::unordered_map<std::string, std::unordered_set<std::string>> label_index_;
  std::unordered_map<std::string, std::unordered_set<std::string>> owner_index_;
};

}  // namespace

PYBIND11_MODULE(engine, m) {
  m.doc() = "kuberay-amd native object store (C++ blobs + indexes + pod views)";

  py::class_<PodViewData>(m, "PodViewData")
      .def(py::init<>())
      .def_readwrite("name", &PodViewData::name)
      .def_readwrite("ns", &PodViewData::ns)
      .def_readwrite("phase", &PodViewData::phase)
      .def_readwrite("deletion_ts", &PodViewData::deletion_ts)
      .def_readwrite("pod_ip", &PodViewData::pod_ip)
      .def_readwrite("restart_policy", &PodViewData::restart_policy)
      .def_readwrite("creation_ts", &PodViewData::creation_ts)
      .def_readwrite("ready", &PodViewData::ready)
      .def_readwrite("terminated", &PodViewData::terminated);

  py::class_<NativeStore>(m, "NativeStore")
      .def(py::init<>())
      .def("put", &NativeStore::put, py::arg("kind"), py::arg("ns"),
           py::arg("name"), py::arg("blob"), py::arg("rv"), py::arg("labels"),
           py::arg("owner_uids"), py::arg("view") = std::nullopt)
      .def("fetch", &NativeStore::fetch)
      .def("rv", &NativeStore::rv)
      .def("contains", &NativeStore::contains)
      .def("remove", &NativeStore::remove)
      .def("list_blobs", &NativeStore::list_blobs, py::arg("kind"),
           py::arg("ns") = std::nullopt,
           py::arg("selector") = std::vector<std::pair<std::string, std::string>>())
      .def("list_views", &NativeStore::list_views, py::arg("ns") = std::nullopt,
           py::arg("selector") = std::vector<std::pair<std::string, std::string>>())
      .def("dependents", &NativeStore::dependents)
      .def("drop_owner", &NativeStore::drop_owner)
      .def("count", &NativeStore::count)
      .def("total_bytes", &NativeStore::total_bytes);
}
