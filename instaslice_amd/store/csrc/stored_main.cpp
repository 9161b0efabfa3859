// instaslice-stored: native store daemon for the instaslice-amd control
// plane.
//
// Speaks exactly the protocol of store/netstore.py (4-byte big-endian length
// + msgpack map per message; verbs create/get/list/update/delete/patch/
// batch/watch/ping with quiet + watch filters + replay) against the object
// semantics of store/memstore.py (resourceVersion optimistic concurrency,
// finalizer/deletionTimestamp two-phase delete, PATCH op grammar, filtered
// watches). The Python MemStore remains the reference implementation — the
// CPU test tier runs the same battery against both (tests/test_native_store
// .py) — but under load the daemon is the control plane's scale-out floor:
// the Python server serializes ~1 ms of GIL-bound wire handling per pod
// lifecycle, which caps an 8-agent cluster near ~400 pods/s; this daemon
// carries the same topology at 3.3-4.2k pods/s (w8, 256-core box, r2).
//
// Throughput discipline (the serialized per-lifecycle event/request chain
// bounds cluster-wide rate; stored itself measures ~3.8 cores at w8 —
// profiles/stored_cpu_w8*.log):
//   - objects are IMMUTABLE once stored (shared_ptr<const Value>); writers
//     build replacements with copy-on-write along the touched path only —
//     a patch clones ~hundreds of bytes of a ~3 KB CR, the rest is shared
//   - watch notification under the lock is a pointer push into each
//     matching connection's outbox; msgpack encoding happens in the
//     per-connection writer threads, outside the lock, in parallel
//   - reads (get/list) return shared references; responses are packed by
//     the reader thread after the lock is released
//
// Reference analog: none — the reference (project-codeflare/instaslice)
// outsources state to the Kubernetes API server + etcd. This daemon is the
// API-server double for standalone/bench deployments.

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdio>
#include <csignal>
#include <cstring>
#include <deque>
#include <tuple>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "msgpack_value.hpp"

namespace stored {

// ---- store errors ----------------------------------------------------------

struct StoreError {
  const char* type;  // "Conflict" | "NotFound" | "AlreadyExists" | "Error"
  std::string msg;
};

// ---- watch plumbing --------------------------------------------------------

struct Conn;  // fwd

struct WatchSub {
  Conn* conn;
  int64_t watch_id;
  bool has_filters = false;
  Value filters;           // array of {kind,name,namespace,labels} maps
  std::string kind;        // kind-only subscription ("" = all)
  std::atomic<bool> dead{false};
};

// One watch event shared by every subscribed connection. The msgpack body
// `[type, obj]` is encoded EXACTLY ONCE (first writer thread to flush it),
// then every connection splices its own {watch_id, event} header around
// the shared bytes — at w8 each CR event fans out to 8+ kind-wide watches
// and per-connection re-encoding of the same ~KB object was the store
// daemon's dominant CPU (measured ~3.6 cores; profiles/stored_cpu_w8.log).
struct EventBlob {
  std::string ev_type;
  std::shared_ptr<const Value> obj;
  std::once_flag once;
  std::string body;  // msgpack [type, obj]

  EventBlob(const char* t, std::shared_ptr<const Value> o)
      : ev_type(t), obj(std::move(o)) {}

  const std::string& ensure_encoded() {
    std::call_once(once, [this] {
      body.push_back(static_cast<char>(0x92));
      pack_str(ev_type, body);
      pack(*obj, body);
    });
    return body;
  }
};

// per-connection outbox: responses and watch events share one ordered queue
// flushed by a dedicated writer thread, so a slow consumer never blocks the
// store mutex or another connection. Watch events enqueue as (watch_id,
// shared EventBlob) and are packed by the writer thread.
struct OutItem {
  std::string bytes;               // pre-framed (responses)
  int64_t wid = -1;                // >= 0: pack {watch_id, event} lazily
  std::shared_ptr<EventBlob> blob;
};

struct Conn {
  int fd;
  std::mutex out_mu;
  std::condition_variable out_cv;
  std::deque<OutItem> outbox;
  bool closing = false;
  std::vector<std::shared_ptr<WatchSub>> subs;
  // serializes actual socket writes between the writer thread and the
  // reader's inline response path (send_now) — byte interleave guard
  std::mutex write_mu;

  bool raw_send(const std::string& data) {
    std::lock_guard<std::mutex> g(write_mu);
    const char* p = data.data();
    size_t left = data.size();
    while (left > 0) {
      ssize_t n = ::send(fd, p, left, MSG_NOSIGNAL);
      if (n <= 0) return false;
      p += n;
      left -= static_cast<size_t>(n);
    }
    return true;
  }

  // responses skip the outbox + writer wakeup when nothing is queued —
  // one fewer thread handoff on every request's critical path. A response
  // may overtake QUEUED watch events; the protocol tolerates that
  // (responses and events resolve independently client-side; only
  // event-vs-event order matters, and that stays in the outbox).
  void send_now(const std::string& frame_bytes) {
    {
      std::lock_guard<std::mutex> g(out_mu);
      if (closing) return;
    }
    if (!raw_send(frame_bytes)) {
      std::lock_guard<std::mutex> g(out_mu);
      closing = true;
      outbox.clear();
      out_cv.notify_one();
    }
  }

  void enqueue(std::string frame) {
    {
      std::lock_guard<std::mutex> g(out_mu);
      if (closing) return;
      OutItem it;
      it.bytes = std::move(frame);
      outbox.push_back(std::move(it));
    }
    out_cv.notify_one();
  }

  void enqueue_event(int64_t wid, std::shared_ptr<EventBlob> blob) {
    {
      std::lock_guard<std::mutex> g(out_mu);
      if (closing) return;
      OutItem it;
      it.wid = wid;
      it.blob = std::move(blob);
      outbox.push_back(std::move(it));
    }
    out_cv.notify_one();
  }

  // push without waking the writer: callers holding the store mutex defer
  // the futex wake until after they release it (notify_one is a syscall;
  // at ~200k event pushes/s it is measurable critical-section time).
  // Slow-consumer guard: a client that stops reading would grow its outbox
  // without bound — past the cap the connection is closed (the k8s API
  // server likewise terminates watchers that can't keep up; a
  // reconnect-enabled client resyncs with replay).
  static constexpr size_t kMaxOutbox = 65536;

  void enqueue_event_silent(int64_t wid, std::shared_ptr<EventBlob> blob) {
    std::lock_guard<std::mutex> g(out_mu);
    if (closing) return;
    if (outbox.size() >= kMaxOutbox) {
      closing = true;
      outbox.clear();
      ::shutdown(fd, SHUT_RDWR);  // unblocks reader+writer; conn tears down
      return;
    }
    OutItem it;
    it.wid = wid;
    it.blob = std::move(blob);
    outbox.push_back(std::move(it));
  }
};

inline std::string frame(const std::string& payload) {
  std::string out;
  out.reserve(payload.size() + 4);
  uint32_t n = static_cast<uint32_t>(payload.size());
  out.push_back(static_cast<char>(n >> 24));
  out.push_back(static_cast<char>(n >> 16));
  out.push_back(static_cast<char>(n >> 8));
  out.push_back(static_cast<char>(n));
  out += payload;
  return out;
}

// ---- copy-on-write helpers -------------------------------------------------

// make the map container at `v` exclusively owned (clone if shared); entries
// keep sharing THEIR children — only the one spine level is copied
inline void cow_map(Value& v) {
  if (v.t != Value::T::Map || !v.m) {
    v = Value::map();
    return;
  }
  if (v.m.use_count() > 1) v.m = std::make_shared<Map>(*v.m);
}

// ---- object store ----------------------------------------------------------

using ObjPtr = std::shared_ptr<const Value>;

class Store {
 public:
  // key = kind \x00 ns \x00 name (lexicographic == Python tuple sort)
  static std::string key(const std::string& kind, const std::string& ns,
                         const std::string& name) {
    std::string k;
    k.reserve(kind.size() + ns.size() + name.size() + 2);
    k += kind;
    k.push_back('\0');
    k += ns;
    k.push_back('\0');
    k += name;
    return k;
  }

  static std::string obj_key(const Value& obj) {
    const Value* md = obj.find("metadata");
    static const std::string kEmpty;
    const std::string& kind = obj.str_or("kind", kEmpty);
    const std::string& ns = md ? md->str_or("namespace", kEmpty) : kEmpty;
    const std::string& name = md ? md->str_or("name", kEmpty) : kEmpty;
    return key(kind, ns, name);
  }

  // `obj` is moved in: the reader thread owns the freshly unpacked request,
  // so no defensive copy is needed before mutation
  Value create(Value&& obj) {
    std::vector<Conn*> wake;
    Value out;
    {
      std::lock_guard<std::mutex> g(mu_);
      std::string k = obj_key(obj);
      if (objects_.count(k))
        throw StoreError{"AlreadyExists", k + " already exists"};
      bump_rv(obj);
      ObjPtr stored = std::make_shared<const Value>(std::move(obj));
      objects_[k] = stored;
      notify_locked("ADDED", stored, wake);
      out = *stored;
    }
    for (Conn* c : wake) c->out_cv.notify_one();
    mark_dirty();
    return out;
  }

  Value get(const std::string& kind, const std::string& ns, const std::string& name) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = objects_.find(key(kind, ns, name));
    if (it == objects_.end())
      throw StoreError{"NotFound", "(" + kind + ", " + ns + ", " + name + ") not found"};
    return *it->second;
  }

  Value list(const std::string& kind, const Value* ns_filter) {
    std::lock_guard<std::mutex> g(mu_);
    Value out = Value::arr();
    std::string prefix = kind;
    prefix.push_back('\0');
    for (auto it = objects_.lower_bound(prefix); it != objects_.end(); ++it) {
      if (it->first.compare(0, prefix.size(), prefix) != 0) break;
      if (ns_filter && ns_filter->t == Value::T::Str) {
        const Value* md = it->second->find("metadata");
        static const std::string kEmpty;
        if (!md || md->str_or("namespace", kEmpty) != ns_filter->s) continue;
      }
      out.a->push_back(*it->second);
    }
    return out;
  }

  Value update(Value&& obj) {
    std::vector<Conn*> wake;
    Value out;
    {
    std::lock_guard<std::mutex> g(mu_);
    std::string k = obj_key(obj);
    auto it = objects_.find(k);
    if (it == objects_.end()) throw StoreError{"NotFound", k + " not found"};
    const Value& cur = *it->second;
    const Value* cur_md = cur.find("metadata");
    const Value* new_md = obj.find("metadata");
    const Value* sent_rv = new_md ? new_md->find("resourceVersion") : nullptr;
    if (sent_rv && !sent_rv->is_nil()) {
      const Value* cur_rv = cur_md ? cur_md->find("resourceVersion") : nullptr;
      if (!cur_rv || !deep_equal(*sent_rv, *cur_rv))
        throw StoreError{"Conflict", k + ": stale resourceVersion"};
    }
    bump_rv(obj);
    // deletionTimestamp is sticky (k8s semantics)
    const Value* cur_dt = cur_md ? cur_md->find("deletionTimestamp") : nullptr;
    if (cur_dt && cur_dt->truthy()) {
      Value* md = obj.find("metadata");
      const Value* new_dt = md ? md->find("deletionTimestamp") : nullptr;
      if (!new_dt || !new_dt->truthy()) md->setkey("deletionTimestamp", *cur_dt);
    }
    out = commit_locked(k, std::move(obj), wake);
    }
    for (Conn* c : wake) c->out_cv.notify_one();
    mark_dirty();
    return out;
  }

  void del(const std::string& kind, const std::string& ns, const std::string& name) {
    std::vector<Conn*> wake;
    {
    std::lock_guard<std::mutex> g(mu_);
    std::string k = key(kind, ns, name);
    auto it = objects_.find(k);
    if (it == objects_.end())
      throw StoreError{"NotFound", "(" + kind + ", " + ns + ", " + name + ") not found"};
    ObjPtr stored = it->second;
    const Value* md = stored->find("metadata");
    const Value* fin = md ? md->find("finalizers") : nullptr;
    if (fin && fin->truthy()) {
      const Value* dt = md->find("deletionTimestamp");
      if (!dt || !dt->truthy()) {
        Value obj = *stored;  // shallow; COW the metadata spine
        cow_map(obj);
        Value* m = obj.find("metadata");
        cow_map(*m);
        double now = std::chrono::duration<double>(
                         std::chrono::system_clock::now().time_since_epoch())
                         .count();
        m->setkey("deletionTimestamp", Value::real(now));
        bump_rv(obj);
        ObjPtr repl = std::make_shared<const Value>(std::move(obj));
        objects_[k] = repl;
        notify_locked("MODIFIED", repl, wake);
      }
    } else {
      // DELETED gets its own fresh rv: every event carries a UNIQUE
      // monotone resourceVersion — the watch resume token (memstore parity)
      Value obj = *stored;
      bump_rv(obj);
      ObjPtr gone = std::make_shared<const Value>(std::move(obj));
      objects_.erase(it);
      notify_locked("DELETED", gone, wake);
    }
    }
    for (Conn* c : wake) c->out_cv.notify_one();
    mark_dirty();
  }

  Value patch(const std::string& kind, const std::string& ns,
              const std::string& name, const Value& ops) {
    std::vector<Conn*> wake;
    Value out;
    {
      std::lock_guard<std::mutex> g(mu_);
      std::string k = key(kind, ns, name);
      auto it = objects_.find(k);
      if (it == objects_.end())
        throw StoreError{"NotFound", "(" + kind + ", " + ns + ", " + name + ") not found"};
      // COW root: shares every container with the stored object until an op
      // touches it. A failed test throws before commit -> stored untouched.
      Value obj = *it->second;
      if (ops.is_arr())
        for (const auto& op : *ops.a) apply_op(obj, op);
      bump_rv(obj);
      out = commit_locked(k, std::move(obj), wake);
    }
    for (Conn* c : wake) c->out_cv.notify_one();
    mark_dirty();
    return out;
  }

  // watch registration: replay/resume + subscribe atomically under the
  // store lock. `respond` is invoked UNDER the lock, before any event is
  // enqueued, so the subscribe response (with rev/resumed) precedes every
  // event on the wire. With has_since and a covered history window, only
  // the missed events replay (resume tokens); a
  // compacted window falls back to the full ADDED relist.
  template <typename RespondFn>
  void add_watch(std::shared_ptr<WatchSub> sub, bool replay,
                 bool has_since, uint64_t since, RespondFn respond) {
    Conn* conn = sub->conn;
    bool pushed = false;
    {
      std::lock_guard<std::mutex> g(mu_);
      bool resumed = false;
      if (has_since) {
        bool covered = history_.empty()
                           ? (since >= rv_)
                           : (since + 1 >= std::get<0>(history_.front()));
        resumed = covered;
      }
      respond(rv_, resumed);
      if (resumed) {
        for (const auto& rec : history_) {
          if (std::get<0>(rec) > since && matches(*sub, *std::get<2>(rec))) {
            conn->enqueue_event_silent(
                sub->watch_id,
                std::make_shared<EventBlob>(std::get<1>(rec).c_str(),
                                            std::get<2>(rec)));
            pushed = true;
          }
        }
      } else if (replay) {
        for (const auto& kv : objects_) {
          if (matches(*sub, *kv.second)) {
            conn->enqueue_event_silent(
                sub->watch_id,
                std::make_shared<EventBlob>("ADDED", kv.second));
            pushed = true;
          }
        }
      }
      watches_.push_back(std::move(sub));
    }
    if (pushed) conn->out_cv.notify_one();
  }

  void drop_conn_watches(Conn* c) {
    std::lock_guard<std::mutex> g(mu_);
    watches_.erase(
        std::remove_if(watches_.begin(), watches_.end(),
                       [c](const std::shared_ptr<WatchSub>& w) {
                         return w->conn == c || w->dead.load();
                       }),
        watches_.end());
  }

 private:
  void bump_rv(Value& obj) {
    ++rv_;
    cow_map(obj);
    Value* md = obj.find("metadata");
    if (!md) md = &obj.setkey("metadata", Value::map());
    cow_map(*md);
    md->setkey("resourceVersion", Value::str(std::to_string(rv_)));
  }

  // shared commit tail for update/patch: store, handle finalizer-free
  // deletion, notify
  Value commit_locked(const std::string& k, Value obj,
                      std::vector<Conn*>& wake) {
    ObjPtr stored = std::make_shared<const Value>(std::move(obj));
    const Value* md = stored->find("metadata");
    const Value* dt = md ? md->find("deletionTimestamp") : nullptr;
    const Value* fin = md ? md->find("finalizers") : nullptr;
    if (dt && dt->truthy() && (!fin || !fin->truthy())) {
      objects_.erase(k);
      notify_locked("DELETED", stored, wake);
    } else {
      objects_[k] = stored;
      notify_locked("MODIFIED", stored, wake);
    }
    return *stored;
  }

  // PATCH op grammar — mirror of memstore.apply_patch_ops, with COW cloning
  // along the touched path only
  void apply_op(Value& obj, const Value& op) {
    const Value* pathv = op.find("path");
    const Value* opname = op.find("op");
    if (!pathv || !pathv->is_arr() || !opname || !opname->is_str())
      throw StoreError{"Error", "malformed patch op"};
    const auto& path = *pathv->a;
    const std::string& kind_op = opname->s;

    if (kind_op == "test") {  // read-only: no COW
      const Value* node = &obj;
      bool missing = false;
      for (const auto& pseg : path) {
        if (!pseg.is_str() || !node->is_map()) { missing = true; break; }
        node = node->find(pseg.s);
        if (!node) { missing = true; break; }
      }
      const Value* absent = op.find("absent");
      if (absent && absent->truthy()) {
        if (!missing) throw StoreError{"Conflict", "patch test: expected absent"};
      } else {
        const Value* want = op.find("value");
        static const Value kNil;
        if (missing || !deep_equal(*node, want ? *want : kNil))
          throw StoreError{"Conflict", "patch test: value mismatch"};
      }
      return;
    }

    if (path.empty()) throw StoreError{"Error", "empty patch path"};
    // navigate to parent, COW-cloning each spine level; lists are leaf
    // containers only — traversing into one errors (MemStore parity,
    // pinned by the differential fuzz test)
    Value* node = &obj;
    cow_map(*node);
    for (size_t k = 0; k + 1 < path.size(); ++k) {
      if (!path[k].is_str()) throw StoreError{"Error", "non-string path"};
      Value* nxt = node->find(path[k].s);
      if (nxt && nxt->is_arr())
        throw StoreError{"Error", "patch path traverses a list"};
      if (!nxt || !nxt->is_map())
        nxt = &node->setkey(path[k].s, Value::map());
      cow_map(*nxt);
      node = nxt;
    }
    if (!path.back().is_str()) throw StoreError{"Error", "non-string path leaf"};
    const std::string& leaf = path.back().s;
    const Value* val = op.find("value");
    static const Value kNil;
    const Value& v = val ? *val : kNil;

    if (kind_op == "set") {
      node->setkey(leaf, v);  // shares request containers; request dies, ptr lives
    } else if (kind_op == "merge") {
      Value* tgt = node->find(leaf);
      if (!tgt || !tgt->is_map()) {
        tgt = &node->setkey(leaf, Value::map());
      } else {
        cow_map(*tgt);
      }
      if (v.is_map())
        for (const auto& kv : *v.m) tgt->setkey(kv.first, kv.second);
    } else if (kind_op == "delete") {
      node->erase(leaf);
    } else if (kind_op == "add_to_set") {
      Value* cur = node->find(leaf);
      Value list = Value::arr();
      if (cur && cur->is_arr()) *list.a = *cur->a;  // element-shallow clone
      bool present = false;
      for (const auto& e : *list.a)
        if (deep_equal(e, v)) { present = true; break; }
      if (!present) {
        list.a->push_back(v);
        std::sort(list.a->begin(), list.a->end(), [](const Value& x, const Value& y) {
          if (x.t == Value::T::Int && y.t == Value::T::Int) return x.i < y.i;
          if (x.is_str() && y.is_str()) return x.s < y.s;
          double xv = x.t == Value::T::Int ? static_cast<double>(x.i) : x.f;
          double yv = y.t == Value::T::Int ? static_cast<double>(y.i) : y.f;
          return xv < yv;
        });
      }
      node->setkey(leaf, std::move(list));
    } else if (kind_op == "remove_from_set") {
      Value* cur = node->find(leaf);
      if (cur && cur->is_arr()) {
        Value list = Value::arr();
        for (const auto& e : *cur->a)
          if (!deep_equal(e, v)) list.a->push_back(e);
        node->setkey(leaf, std::move(list));
      }
    } else if (kind_op == "delete_where") {
      // predicate delete on the FRESH object (mirror of memstore):
      // drop map entries whose entry[field] == value
      const Value* fieldv = op.find("field");
      Value* cur = node->find(leaf);
      if (fieldv && fieldv->is_str() && cur && cur->is_map()) {
        Value kept = Value::map();
        for (const auto& kv : *cur->m) {
          const Value* got = kv.second.is_map() ? kv.second.find(fieldv->s)
                                                : nullptr;
          if (!(got && deep_equal(*got, v))) kept.setkey(kv.first, kv.second);
        }
        node->setkey(leaf, std::move(kept));
      }
    } else {
      throw StoreError{"Error", "unknown patch op " + kind_op};
    }
  }

  static bool filter_matches(const Value& f, const Value& obj) {
    static const std::string kEmpty;
    const Value* want_kind = f.find("kind");
    if (want_kind && want_kind->is_str() &&
        want_kind->s != obj.str_or("kind", kEmpty))
      return false;
    const Value* md = obj.find("metadata");
    const Value* want_name = f.find("name");
    if (want_name && want_name->is_str()) {
      if (!md || md->str_or("name", kEmpty) != want_name->s) return false;
    }
    const Value* want_ns = f.find("namespace");
    if (want_ns && want_ns->is_str()) {
      if (!md || md->str_or("namespace", kEmpty) != want_ns->s) return false;
    }
    const Value* want_labels = f.find("labels");
    if (want_labels && want_labels->is_map() && !want_labels->m->empty()) {
      const Value* have = md ? md->find("labels") : nullptr;
      for (const auto& kv : *want_labels->m) {
        const Value* hv = have ? have->find(kv.first) : nullptr;
        if (!hv || !deep_equal(*hv, kv.second)) return false;
      }
    }
    return true;
  }

  static bool matches(const WatchSub& w, const Value& obj) {
    if (w.has_filters) {
      for (const auto& f : *w.filters.a)
        if (filter_matches(f, obj)) return true;
      return false;
    }
    static const std::string kEmpty;
    return w.kind.empty() || w.kind == obj.str_or("kind", kEmpty);
  }

  // under the lock: pointer pushes only — packing happens in writer threads
  // and the futex wakes are deferred to after the store mutex is released
  void notify_locked(const char* type, const ObjPtr& obj,
                     std::vector<Conn*>& wake) {
    // bounded event history for watch resume (rv_ was bumped by the
    // mutation that triggers this notify, so it is the event's revision)
    history_.emplace_back(rv_, std::string(type), obj);
    if (history_.size() > kHistoryMax) history_.pop_front();
    bool any_dead = false;
    // ONE blob per event: every matching connection shares the encoded body
    std::shared_ptr<EventBlob> blob;
    for (const auto& w : watches_) {
      if (w->dead.load()) { any_dead = true; continue; }
      if (!matches(*w, *obj)) continue;
      if (!blob) blob = std::make_shared<EventBlob>(type, obj);
      w->conn->enqueue_event_silent(w->watch_id, blob);
      if (wake.empty() || wake.back() != w->conn) wake.push_back(w->conn);
    }
    if (any_dead) {
      watches_.erase(std::remove_if(watches_.begin(), watches_.end(),
                                    [](const std::shared_ptr<WatchSub>& w) {
                                      return w->dead.load();
                                    }),
                     watches_.end());
    }
  }

  std::mutex mu_;
  std::map<std::string, ObjPtr> objects_;
  uint64_t rv_ = 0;
  static constexpr size_t kHistoryMax = 8192;
  std::deque<std::tuple<uint64_t, std::string, ObjPtr>> history_;
  std::vector<std::shared_ptr<WatchSub>> watches_;

  // -- persistence (checkpoint/resume; msgpack snapshot, write-behind) ----
  // Python MemStore snapshots JSON; the daemon snapshots the same
  // {"rv": N, "objects": [...]} shape in msgpack. Debounced: bursts of
  // mutations coalesce into one write, and SIGTERM flushes synchronously.
  std::string persist_path_;
  std::atomic<bool> persist_dirty_{false};

 public:
  void set_persist_path(std::string path) { persist_path_ = std::move(path); }

  void mark_dirty() {
    if (!persist_path_.empty()) persist_dirty_.store(true);
  }

  bool load_snapshot() {
    if (persist_path_.empty()) return false;
    FILE* f = std::fopen(persist_path_.c_str(), "rb");
    if (!f) return false;
    std::string buf;
    char tmp[1 << 16];
    size_t n;
    while ((n = std::fread(tmp, 1, sizeof(tmp), f)) > 0) buf.append(tmp, n);
    std::fclose(f);
    try {
      Value snap = unpack(buf);
      const Value* rv = snap.find("rv");
      const Value* objs = snap.find("objects");
      std::lock_guard<std::mutex> g(mu_);
      if (rv && rv->t == Value::T::Int) rv_ = static_cast<uint64_t>(rv->i);
      if (objs && objs->is_arr())
        for (const auto& o : *objs->a)
          objects_[obj_key(o)] = std::make_shared<const Value>(o);
      return true;
    } catch (const std::exception& e) {
      std::fprintf(stderr, "snapshot load failed: %s\n", e.what());
      return false;
    }
  }

  void flush_snapshot() {
    if (persist_path_.empty()) return;
    Value snap = Value::map();
    {
      std::lock_guard<std::mutex> g(mu_);
      snap.setkey("rv", Value::integer(static_cast<int64_t>(rv_)));
      Value arr = Value::arr();
      arr.a->reserve(objects_.size());
      for (const auto& kv : objects_) arr.a->push_back(*kv.second);
      snap.setkey("objects", std::move(arr));
    }
    std::string payload;  // packed OUTSIDE the lock (objects are immutable)
    pack(snap, payload);
    std::string tmp_path = persist_path_ + ".tmp";
    FILE* f = std::fopen(tmp_path.c_str(), "wb");
    if (!f) return;
    std::fwrite(payload.data(), 1, payload.size(), f);
    std::fclose(f);
    std::rename(tmp_path.c_str(), persist_path_.c_str());
  }

  bool take_dirty() { return persist_dirty_.exchange(false); }
};

// ---- per-connection handling ----------------------------------------------

Value ok_response(const Value* rid, Value result) {
  Value resp = Value::map();
  if (rid) resp.setkey("id", *rid);
  resp.setkey("ok", Value::boolean(true));
  resp.setkey("result", std::move(result));
  return resp;
}

Value err_response(const Value* rid, const StoreError& e) {
  Value resp = Value::map();
  if (rid) resp.setkey("id", *rid);
  resp.setkey("ok", Value::boolean(false));
  Value err = Value::map();
  err.setkey("type", Value::str(e.type));
  err.setkey("msg", Value::str(e.msg));
  resp.setkey("error", std::move(err));
  return resp;
}

// `req` is mutable: create/update MOVE the obj payload into the store
Value execute(Store& store, Value& req) {
  static const std::string kEmpty;
  const std::string& verb = req.str_or("verb", kEmpty);
  const std::string& kind = req.str_or("kind", kEmpty);
  const std::string& name = req.str_or("name", kEmpty);
  const std::string& ns = req.str_or("namespace", kEmpty);
  if (verb == "create") {
    Value* obj = req.find("obj");
    if (!obj) throw StoreError{"Error", "create: missing obj"};
    return store.create(std::move(*obj));
  }
  if (verb == "get") return store.get(kind, ns, name);
  if (verb == "list") return store.list(kind, req.find("namespace"));
  if (verb == "update") {
    Value* obj = req.find("obj");
    if (!obj) throw StoreError{"Error", "update: missing obj"};
    return store.update(std::move(*obj));
  }
  if (verb == "delete") {
    store.del(kind, ns, name);
    return Value::nil();
  }
  if (verb == "patch") {
    const Value* ops = req.find("ops");
    static const Value kNilOps;
    return store.patch(kind, ns, name, ops ? *ops : kNilOps);
  }
  if (verb == "ping") return Value::str("pong");
  throw StoreError{"Error", "unknown verb '" + verb + "'"};
}

// pack one watch event: {"watch_id": N, "event": [type, obj]} — the event
// body bytes are encoded once per EVENT (EventBlob), shared across every
// subscribed connection; only the tiny per-watch header is built here.
std::string pack_event(const OutItem& it) {
  const std::string& body = it.blob->ensure_encoded();
  std::string payload;
  payload.reserve(body.size() + 24);
  payload.push_back(static_cast<char>(0x82));
  pack_str("watch_id", payload);
  pack_int(it.wid, payload);
  pack_str("event", payload);
  payload += body;
  return frame(payload);
}

void writer_loop(std::shared_ptr<Conn> conn) {
  for (;;) {
    std::deque<OutItem> batch;
    {
      std::unique_lock<std::mutex> g(conn->out_mu);
      conn->out_cv.wait(g, [&] { return conn->closing || !conn->outbox.empty(); });
      if (conn->closing && conn->outbox.empty()) return;
      batch.swap(conn->outbox);
    }
    std::string data;
    for (auto& it : batch)
      data += (it.wid >= 0) ? pack_event(it) : std::move(it.bytes);
    if (!conn->raw_send(data)) {
      std::lock_guard<std::mutex> g(conn->out_mu);
      conn->closing = true;
      conn->outbox.clear();
      return;
    }
  }
}

bool read_exact(int fd, char* buf, size_t n) {
  size_t got = 0;
  while (got < n) {
    ssize_t r = ::recv(fd, buf + got, n - got, 0);
    if (r <= 0) return false;
    got += static_cast<size_t>(r);
  }
  return true;
}

void reader_loop(Store& store, std::shared_ptr<Conn> conn) {
  std::thread writer(writer_loop, conn);
  int64_t next_watch_id = 0;
  for (;;) {
    char hdr[4];
    if (!read_exact(conn->fd, hdr, 4)) break;
    uint32_t len = (static_cast<uint8_t>(hdr[0]) << 24) |
                   (static_cast<uint8_t>(hdr[1]) << 16) |
                   (static_cast<uint8_t>(hdr[2]) << 8) |
                   static_cast<uint8_t>(hdr[3]);
    if (len > (64u << 20)) break;  // oversized frame: drop connection
    std::string buf(len, '\0');
    if (!read_exact(conn->fd, buf.data(), len)) break;

    Value req;
    try {
      req = unpack(buf);
    } catch (const std::exception&) {
      break;  // protocol desync: close
    }
    const Value* rid = req.find("id");
    static const std::string kEmpty;
    const std::string& verb = req.str_or("verb", kEmpty);
    const Value* quietv = req.find("quiet");
    bool quiet = quietv && quietv->truthy();

    Value resp;
    try {
      if (verb == "watch") {
        auto sub = std::make_shared<WatchSub>();
        sub->conn = conn.get();
        sub->watch_id = ++next_watch_id;
        const Value* filters = req.find("filters");
        if (filters && filters->is_arr()) {
          sub->has_filters = true;
          sub->filters = *filters;
        } else {
          const Value* kindv = req.find("kind");
          sub->kind = (kindv && kindv->is_str()) ? kindv->s : "";
        }
        const Value* replayv = req.find("replay");
        bool replay = !replayv || replayv->truthy();
        const Value* sincev = req.find("since");
        bool has_since = sincev && sincev->t == Value::T::Int &&
                         sincev->i >= 0;
        uint64_t since = has_since ? static_cast<uint64_t>(sincev->i) : 0;
        conn->subs.push_back(sub);
        int64_t wid = sub->watch_id;
        Conn* cptr = conn.get();
        // respond inside add_watch's lock, before any event is enqueued:
        // the watch_id + rev/resumed precede every event on the wire
        store.add_watch(sub, replay, has_since, since,
                        [&](uint64_t rev, bool resumed) {
          Value result = Value::map();
          result.setkey("watch_id", Value::integer(wid));
          result.setkey("rev", Value::integer(static_cast<int64_t>(rev)));
          result.setkey("resumed", Value::boolean(resumed));
          std::string payload;
          pack(ok_response(rid, std::move(result)), payload);
          cptr->send_now(frame(payload));
        });
        continue;
      }
      if (verb == "batch") {
        Value* reqs = req.find("requests");
        Value results = Value::arr();
        if (reqs && reqs->is_arr()) {
          for (auto& sub_req : *reqs->a) {
            Value entry = Value::map();
            try {
              Value r = execute(store, sub_req);
              entry.setkey("ok", Value::boolean(true));
              entry.setkey("result", quiet ? Value::nil() : std::move(r));
            } catch (const StoreError& e) {
              entry.setkey("ok", Value::boolean(false));
              entry.setkey("result", Value::nil());
              Value err = Value::map();
              err.setkey("type", Value::str(e.type));
              err.setkey("msg", Value::str(e.msg));
              entry.setkey("error", std::move(err));
            }
            results.a->push_back(std::move(entry));
          }
        }
        resp = ok_response(rid, std::move(results));
      } else {
        Value r = execute(store, req);
        resp = ok_response(rid, quiet ? Value::nil() : std::move(r));
      }
    } catch (const StoreError& e) {
      resp = err_response(rid, e);
    } catch (const std::exception& e) {
      resp = err_response(rid, StoreError{"Error", e.what()});
    }
    std::string payload;
    pack(resp, payload);
    conn->send_now(frame(payload));
  }

  // teardown
  for (auto& s : conn->subs) s->dead.store(true);
  store.drop_conn_watches(conn.get());
  {
    std::lock_guard<std::mutex> g(conn->out_mu);
    conn->closing = true;
  }
  conn->out_cv.notify_one();
  writer.join();
  ::close(conn->fd);
}

}  // namespace stored

static std::atomic<bool> g_stop{false};

static void on_signal(int) { g_stop.store(true); }

int main(int argc, char** argv) {
  using namespace stored;
  int port = 0;
  if (argc > 1) port = std::atoi(argv[1]);
  const char* snapshot = (argc > 2) ? argv[2] : nullptr;

  int srv = ::socket(AF_INET, SOCK_STREAM, 0);
  if (srv < 0) { perror("socket"); return 1; }
  int one = 1;
  ::setsockopt(srv, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
  addr.sin_port = htons(static_cast<uint16_t>(port));
  if (::bind(srv, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) != 0) {
    perror("bind");
    return 1;
  }
  socklen_t alen = sizeof(addr);
  ::getsockname(srv, reinterpret_cast<sockaddr*>(&addr), &alen);
  if (::listen(srv, 64) != 0) { perror("listen"); return 1; }
  // parent (store/native.py) parses this line for the chosen port
  Store store;
  if (snapshot) {
    store.set_persist_path(snapshot);
    if (store.load_snapshot())
      std::fprintf(stderr, "resumed from %s\n", snapshot);
    std::signal(SIGTERM, on_signal);
    std::signal(SIGINT, on_signal);
    // write-behind snapshot thread (200 ms debounce) + SIGTERM flush
    std::thread([&store] {
      for (;;) {
        std::this_thread::sleep_for(std::chrono::milliseconds(100));
        if (g_stop.load()) {
          store.flush_snapshot();
          std::_Exit(0);
        }
        if (store.take_dirty()) {
          std::this_thread::sleep_for(std::chrono::milliseconds(100));
          store.take_dirty();
          store.flush_snapshot();
        }
      }
    }).detach();
  }
  std::printf("LISTENING %d\n", ntohs(addr.sin_port));
  std::fflush(stdout);

  for (;;) {
    int fd = ::accept(srv, nullptr, nullptr);
    if (fd < 0) {
      if (errno == EINTR) continue;
      break;
    }
    ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    auto conn = std::make_shared<Conn>();
    conn->fd = fd;
    std::thread(reader_loop, std::ref(store), conn).detach();
  }
  return 0;
}
