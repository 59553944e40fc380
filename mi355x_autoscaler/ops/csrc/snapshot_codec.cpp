// High-throughput PodList JSON -> SoA snapshot encoder (C++/pybind11).
//
// Re-creation of the reference's ingest path (pykube JSON -> KubePod
// wrapping [R: autoscaler/kube.py — reconstructed]) as a native codec
// [SURVEY §2.2 row F]: a single pass over the API server's PodList
// body materializes a tiny string_view DOM per item and emits the pod
// columns of snapshot/encode.py's integer lattice. Every rule —
// quantity parsing, MiB round-up, FNV-1a pair hashing, flag bits,
// owner-kind sets — mirrors the Python encoder exactly; the
// differential test (tests/test_codec.py) asserts byte-equality on
// every column.
//
// The Python path is the semantic spec; this is the fast path the
// control loop uses when the extension is built (ctl/loop.py).
#include <torch/extension.h>

#include <pybind11/numpy.h>

#include <algorithm>
#include <cctype>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <atomic>
#include <chrono>
#include <deque>
#include <memory>
#include <mutex>
#include <thread>
#include <stdexcept>
#include <string>
#include <string_view>
#include <unordered_map>
#include <vector>

namespace {

// ---- lattice constants (must match snapshot/encode.py) --------------------
constexpr int NRES = 4;
constexpr int MAXSEL = 32;
// int64 view of uint64 max — mirrors encode.py::SELECTOR_POISON
constexpr int64_t SELECTOR_POISON = -1;
// pod LABEL pair width — mirrors encode.py::MAX_POD_LABEL_PAIRS
// (host-only column: PDB re-matching in the watch cache)
constexpr int MAXPODLBL = 16;
constexpr int64_t MIB = 1024 * 1024;

constexpr uint16_t PF_PENDING = 1 << 0;
constexpr uint16_t PF_MIRROR = 1 << 1;
constexpr uint16_t PF_DAEMONSET = 1 << 2;
constexpr uint16_t PF_CRITICAL = 1 << 3;
constexpr uint16_t PF_REPLICATED = 1 << 4;
constexpr uint16_t PF_DRAINABLE = 1 << 5;
constexpr uint16_t PF_BLOCKS_DRAIN = 1 << 6;
constexpr uint16_t PF_SEL_OVERFLOW = 1 << 7;
constexpr uint16_t PF_BOUND = 1 << 8;

// FNV-1a over key + 0x1f + value, identical to hashing the
// concatenated string (snapshot/encode.py::hash_label_pair).
uint64_t hash_label_pair(std::string_view k, std::string_view v) {
  uint64_t h = 0xCBF29CE484222325ull;
  for (unsigned char c : k) {
    h ^= c;
    h *= 0x100000001B3ull;
  }
  h ^= 0x1f;
  h *= 0x100000001B3ull;
  for (unsigned char c : v) {
    h ^= c;
    h *= 0x100000001B3ull;
  }
  return h;
}

// ---- minimal JSON DOM (zero-copy string_views, pooled nodes) ------------
struct JValue;
using JMembers = std::vector<std::pair<std::string_view, JValue*>>;
using JItems = std::vector<JValue*>;

struct JValue {
  enum Kind { Null, Bool, Num, Str, Obj, Arr } kind = Null;
  bool b = false;
  double num = 0;
  std::string_view str;  // view into the body (or the decode side-buffer)
  JMembers members;      // Obj
  JItems items;          // Arr
};

class Parser {
 public:
  Parser(const char* p, size_t n) : p_(p), begin_(p), end_(p + n) {}
  ~Parser() {
    for (JValue* v : pool_) delete v;
  }

  JValue* parse_value() {
    skip_ws();
    if (p_ >= end_) fail("unexpected end");
    switch (*p_) {
      case '{':
        return parse_object();
      case '[':
        return parse_array();
      case '"': {
        JValue* v = alloc();
        v->kind = JValue::Str;
        v->str = parse_string();
        return v;
      }
      case 't':
        expect("true");
        return make_bool(true);
      case 'f':
        expect("false");
        return make_bool(false);
      case 'n':
        expect("null");
        return alloc();
      default:
        return parse_number();
    }
  }

  void skip_ws() {
    while (p_ < end_ && (*p_ == ' ' || *p_ == '\t' || *p_ == '\n' || *p_ == '\r')) ++p_;
  }

  bool at(char c) {
    skip_ws();
    return p_ < end_ && *p_ == c;
  }

  void consume(char c) {
    skip_ws();
    if (p_ >= end_ || *p_ != c) fail(std::string("expected '") + c + "'");
    ++p_;
  }

  void expect(const char* lit) {
    const size_t n = strlen(lit);
    if ((size_t)(end_ - p_) < n || strncmp(p_, lit, n) != 0) fail("bad literal");
    p_ += n;
  }

  std::string_view parse_string() {
    consume('"');
    const char* start = p_;
    // fast path: no escapes -> zero-copy view into the body
    while (p_ < end_ && *p_ != '"' && *p_ != '\\') ++p_;
    if (p_ < end_ && *p_ == '"') {
      std::string_view sv(start, (size_t)(p_ - start));
      ++p_;
      return sv;
    }
    // slow path: decode escapes into the (address-stable) side buffer
    decoded_.emplace_back(start, (size_t)(p_ - start));
    std::string& out = decoded_.back();
    while (p_ < end_) {
      char c = *p_++;
      if (c == '"') return std::string_view(out);
      if (c == '\\') {
        if (p_ >= end_) break;
        char e = *p_++;
        switch (e) {
          case '"': out += '"'; break;
          case '\\': out += '\\'; break;
          case '/': out += '/'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'n': out += '\n'; break;
          case 'r': out += '\r'; break;
          case 't': out += '\t'; break;
          case 'u': {
            unsigned cp = parse_hex4();
            if (cp >= 0xD800 && cp <= 0xDBFF && end_ - p_ >= 6 && p_[0] == '\\' &&
                p_[1] == 'u') {
              p_ += 2;
              unsigned lo = parse_hex4();
              cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
            }
            if (cp < 0x80) out += (char)cp;
            else if (cp < 0x800) {
              out += (char)(0xC0 | (cp >> 6));
              out += (char)(0x80 | (cp & 0x3F));
            } else if (cp < 0x10000) {
              out += (char)(0xE0 | (cp >> 12));
              out += (char)(0x80 | ((cp >> 6) & 0x3F));
              out += (char)(0x80 | (cp & 0x3F));
            } else {
              out += (char)(0xF0 | (cp >> 18));
              out += (char)(0x80 | ((cp >> 12) & 0x3F));
              out += (char)(0x80 | ((cp >> 6) & 0x3F));
              out += (char)(0x80 | (cp & 0x3F));
            }
            break;
          }
          default:
            fail("bad escape");
        }
      } else {
        out += c;
      }
    }
    fail("unterminated string");
    return {};
  }

  unsigned parse_hex4() {
    if (end_ - p_ < 4) fail("bad \\u escape");
    unsigned cp = 0;
    for (int i = 0; i < 4; ++i) {
      char h = *p_++;
      cp <<= 4;
      if (h >= '0' && h <= '9') cp |= h - '0';
      else if (h >= 'a' && h <= 'f') cp |= h - 'a' + 10;
      else if (h >= 'A' && h <= 'F') cp |= h - 'A' + 10;
      else fail("bad hex");
    }
    return cp;
  }

  JValue* parse_object() {
    consume('{');
    JValue* v = alloc();
    v->kind = JValue::Obj;
    skip_ws();
    if (at('}')) {
      ++p_;
      return v;
    }
    while (true) {
      skip_ws();
      std::string_view key = parse_string();
      consume(':');
      JValue* val = parse_value();
      v->members.emplace_back(key, val);
      skip_ws();
      if (at(',')) {
        ++p_;
        continue;
      }
      consume('}');
      return v;
    }
  }

  JValue* parse_array() {
    consume('[');
    JValue* v = alloc();
    v->kind = JValue::Arr;
    skip_ws();
    if (at(']')) {
      ++p_;
      return v;
    }
    while (true) {
      v->items.push_back(parse_value());
      skip_ws();
      if (at(',')) {
        ++p_;
        continue;
      }
      consume(']');
      return v;
    }
  }

  JValue* parse_number() {
    char buf[64];
    int n = 0;
    while (p_ < end_ && n < 63 &&
           (*p_ == '-' || *p_ == '+' || *p_ == '.' || *p_ == 'e' || *p_ == 'E' ||
            (*p_ >= '0' && *p_ <= '9'))) {
      buf[n++] = *p_++;
    }
    buf[n] = 0;
    JValue* v = alloc();
    v->kind = JValue::Num;
    v->num = strtod(buf, nullptr);
    return v;
  }

  // node pool: reused across items (reset_arena keeps capacity)
  JValue* alloc() {
    if (used_ == pool_.size()) pool_.push_back(new JValue);
    JValue* v = pool_[used_++];
    v->kind = JValue::Null;
    v->b = false;
    v->num = 0;
    v->str = {};
    v->members.clear();
    v->items.clear();
    return v;
  }
  JValue* make_bool(bool b) {
    JValue* v = alloc();
    v->kind = JValue::Bool;
    v->b = b;
    return v;
  }
  void reset_arena() {
    used_ = 0;
    decoded_.clear();
  }

  void reset_to(const char* p, size_t n) {
    p_ = p;
    begin_ = p;
    end_ = p + n;
    reset_arena();
  }

  [[noreturn]] void fail(const std::string& msg) {
    throw std::runtime_error("json parse error at offset " +
                             std::to_string(p_ - begin_) + ": " + msg);
  }

  const char* p_;
  const char* begin_;
  const char* end_;
  std::vector<JValue*> pool_;
  size_t used_ = 0;
  std::deque<std::string> decoded_;
};

const JValue* get(const JValue* obj, std::string_view key) {
  if (!obj || obj->kind != JValue::Obj) return nullptr;
  for (auto& m : obj->members)
    if (m.first == key) return m.second;
  return nullptr;
}

std::string_view get_str(const JValue* obj, std::string_view key) {
  const JValue* v = get(obj, key);
  return v && v->kind == JValue::Str ? v->str : std::string_view();
}

// ---- quantity parsing (mirrors models/resources.py) -----------------------
double parse_quantity(const JValue* v) {
  if (!v) return 0.0;
  if (v->kind == JValue::Num) return v->num;
  if (v->kind != JValue::Str) return 0.0;
  std::string_view s = v->str;
  if (s.empty()) return 0.0;
  auto to_num = [](std::string_view sv) {
    char buf[64];
    const size_t n = std::min(sv.size(), sizeof(buf) - 1);
    memcpy(buf, sv.data(), n);
    buf[n] = 0;
    return strtod(buf, nullptr);
  };
  static const std::pair<const char*, double> kBinary[] = {
      {"Ki", 1024.0}, {"Mi", 1048576.0}, {"Gi", 1073741824.0},
      {"Ti", 1099511627776.0}, {"Pi", 1125899906842624.0},
      {"Ei", 1152921504606846976.0},
  };
  for (auto& [suf, mult] : kBinary) {
    size_t sl = strlen(suf);
    if (s.size() > sl && s.compare(s.size() - sl, sl, suf) == 0) {
      return to_num(s.substr(0, s.size() - sl)) * mult;
    }
  }
  char last = s.back();
  double mult = 0;
  switch (last) {
    case 'n': mult = 1e-9; break;
    case 'u': mult = 1e-6; break;
    case 'm': mult = 1e-3; break;
    case 'k': mult = 1e3; break;
    case 'M': mult = 1e6; break;
    case 'G': mult = 1e9; break;
    case 'T': mult = 1e12; break;
    case 'P': mult = 1e15; break;
    case 'E': mult = 1e18; break;
    default:
      return to_num(s);
  }
  return to_num(s.substr(0, s.size() - 1)) * mult;
}

int64_t parse_cpu_milli(const JValue* v) {
  double f = parse_quantity(v) * 1000.0;
  return f > 0 ? (int64_t)(f + 0.5) : 0;
}

int64_t parse_mem_bytes(const JValue* v) {
  double f = parse_quantity(v);
  return f > 0 ? (int64_t)(f + 0.5) : 0;
}

bool is_gpu_key(std::string_view k) {
  return k == "amd.com/gpu" || k == "alpha.kubernetes.io/nvidia-gpu" ||
         k == "nvidia.com/gpu";
}

struct Req {
  int64_t cpu = 0, mem = 0, pods = 0, gpu = 0;
};

Req sum_requests(const JValue* spec) {
  Req total;
  total.pods = 1;
  const JValue* containers = get(spec, "containers");
  if (containers && containers->kind == JValue::Arr) {
    for (const JValue* c : containers->items) {
      const JValue* req = get(get(c, "resources"), "requests");
      if (!req || req->kind != JValue::Obj) continue;
      for (auto& m : req->members) {
        if (m.first == "cpu") total.cpu += parse_cpu_milli(m.second);
        else if (m.first == "memory") total.mem += parse_mem_bytes(m.second);
        else if (is_gpu_key(m.first)) total.gpu += (int64_t)parse_quantity(m.second);
      }
    }
  }
  const JValue* inits = get(spec, "initContainers");
  if (inits && inits->kind == JValue::Arr) {
    for (const JValue* c : inits->items) {
      const JValue* req = get(get(c, "resources"), "requests");
      if (!req || req->kind != JValue::Obj) continue;
      Req iv;
      for (auto& m : req->members) {
        if (m.first == "cpu") iv.cpu = parse_cpu_milli(m.second);
        else if (m.first == "memory") iv.mem = parse_mem_bytes(m.second);
        else if (is_gpu_key(m.first)) iv.gpu += (int64_t)parse_quantity(m.second);
      }
      total.cpu = std::max(total.cpu, iv.cpu);
      total.mem = std::max(total.mem, iv.mem);
      total.gpu = std::max(total.gpu, iv.gpu);
    }
  }
  return total;
}

bool kind_is_replicated(std::string_view k) {
  return k == "ReplicationController" || k == "ReplicaSet" || k == "StatefulSet" ||
         k == "Job" || k == "DeploymentConfig" || k == "DaemonSet";
}

bool ieq_true(std::string_view s) {
  std::string t;
  for (char c : s) t += (char)tolower((unsigned char)c);
  return t == "true" || t == "'true'";
}

}  // namespace

// ---------------------------------------------------------------------------
// Parallel encode: boundary pre-scan -> per-thread parse (GIL released)
// -> column concatenation.
// ---------------------------------------------------------------------------

struct PdbMatcher {
  std::string ns;  // a v1 PDB selects pods in ITS OWN namespace only
  std::vector<std::pair<std::string, std::string>> labels;
};

struct Ctx {
  std::unordered_map<std::string, int32_t> node_idx;
  std::vector<PdbMatcher> pdbs;
};

// Parse the Python-side matcher list: entries are (namespace, labels)
// tuples, or bare label dicts (legacy — namespace 'default'); mirrors
// snapshot/delta.py::_norm_matchers.
static void parse_pdb_matchers(pybind11::list pdb_selectors, Ctx& ctx) {
  namespace py = pybind11;
  for (auto sel : pdb_selectors) {
    PdbMatcher m;
    py::object labels_obj = py::reinterpret_borrow<py::object>(sel);
    if (py::isinstance<py::tuple>(labels_obj) || py::isinstance<py::list>(labels_obj)) {
      auto seq = labels_obj.cast<py::sequence>();
      m.ns = py::cast<std::string>(seq[0]);
      labels_obj = py::reinterpret_borrow<py::object>(seq[1]);
    } else {
      m.ns = "default";
    }
    for (auto kv : labels_obj.cast<py::dict>())
      m.labels.emplace_back(py::cast<std::string>(kv.first),
                            py::cast<std::string>(kv.second));
    ctx.pdbs.push_back(std::move(m));
  }
}

struct Columns {
  std::vector<int32_t> pod_req;
  std::vector<int32_t> pod_flags;
  std::vector<int64_t> pod_sel_hash;
  std::vector<int64_t> pod_sel_pairs;
  std::vector<int32_t> pod_sel_count;
  std::vector<int32_t> pod_node;
  std::vector<int32_t> pod_pdb;
  std::vector<int64_t> pod_label_pairs;
  std::vector<int32_t> pod_label_count;
  std::vector<std::string> uids, ns, names;

  void reserve(size_t n) {
    pod_req.reserve(n * NRES);
    pod_flags.reserve(n);
    pod_sel_hash.reserve(n);
    pod_sel_pairs.reserve(n * MAXSEL);
    pod_sel_count.reserve(n);
    pod_node.reserve(n);
    pod_pdb.reserve(n);
    pod_label_pairs.reserve(n * MAXPODLBL);
    pod_label_count.reserve(n);
    uids.reserve(n);
    ns.reserve(n);
    names.reserve(n);
  }
};

void encode_one(const JValue* pod, const Ctx& ctx, Columns& out) {
  const JValue* meta = get(pod, "metadata");
  const JValue* spec = get(pod, "spec");
  const JValue* status = get(pod, "status");

  // requests (quantize: cpu milli, memory ceil-MiB)
  Req rq = sum_requests(spec);
  out.pod_req.push_back((int32_t)std::min<int64_t>(rq.cpu, INT32_MAX));
  out.pod_req.push_back((int32_t)std::min<int64_t>((rq.mem + MIB - 1) / MIB, INT32_MAX));
  out.pod_req.push_back((int32_t)rq.pods);
  out.pod_req.push_back((int32_t)std::min<int64_t>(rq.gpu, INT32_MAX));

  // selectors
  const JValue* nsel = get(spec, "nodeSelector");
  std::vector<uint64_t> hashes;
  if (nsel && nsel->kind == JValue::Obj) {
    for (auto& m : nsel->members) {
      if (m.second->kind == JValue::Str)
        hashes.push_back(hash_label_pair(m.first, m.second->str));
    }
  }
  std::sort(hashes.begin(), hashes.end());
  uint64_t combined = 0;
  for (uint64_t h : hashes) combined ^= h;
  int cnt = (int)hashes.size();
  for (int q = 0; q < MAXSEL; ++q)
    out.pod_sel_pairs.push_back(q < cnt ? (int64_t)hashes[q] : 0);
  if (cnt > MAXSEL)  // wider than the lattice: poison -> matches nothing
    out.pod_sel_pairs[out.pod_sel_pairs.size() - 1] = SELECTOR_POISON;
  out.pod_sel_count.push_back((int32_t)std::min(cnt, MAXSEL));
  out.pod_sel_hash.push_back((int64_t)combined);

  // flags
  int32_t f = 0;
  const std::string_view phase = get_str(status, "phase");
  const std::string_view node_name = get_str(spec, "nodeName");
  if (phase == "Pending" && node_name.empty()) {
    const JValue* conds = get(status, "conditions");
    if (conds && conds->kind == JValue::Arr) {
      for (const JValue* c : conds->items) {
        if (get_str(c, "type") == "PodScheduled" &&
            get_str(c, "status") == "False" &&
            get_str(c, "reason") == "Unschedulable") {
          f |= PF_PENDING;
          break;
        }
      }
    }
  }
  bool mirror = false, critical = false, daemonset = false, replicated = false;
  bool evict_opt_out = false;
  const JValue* annotations = get(meta, "annotations");
  if (annotations && annotations->kind == JValue::Obj) {
    for (auto& m : annotations->members) {
      if (m.first == "kubernetes.io/config.mirror") mirror = true;
      else if (m.first == "scheduler.alpha.kubernetes.io/critical-pod" &&
               m.second->kind == JValue::Str && ieq_true(m.second->str))
        critical = true;
      else if (m.first == "cluster-autoscaler.kubernetes.io/safe-to-evict" &&
               m.second->kind == JValue::Str) {
        std::string t;
        for (char ch : m.second->str) t += (char)tolower((unsigned char)ch);
        if (t == "false") evict_opt_out = true;
      }
    }
  }
  const JValue* owners = get(meta, "ownerReferences");
  if (owners && owners->kind == JValue::Arr) {
    for (const JValue* o : owners->items) {
      const std::string_view kind = get_str(o, "kind");
      if (kind == "DaemonSet") daemonset = true;
      if (kind_is_replicated(kind)) replicated = true;
    }
  }
  if (mirror) f |= PF_MIRROR;
  if (daemonset) f |= PF_DAEMONSET;
  if (critical) f |= PF_CRITICAL;
  if (replicated) f |= PF_REPLICATED;
  // drainable / blocks-drain mirror models/kube.py exactly
  bool drainable = !evict_opt_out && (replicated || mirror || daemonset) && !critical;
  if (drainable) f |= PF_DRAINABLE;
  bool blocks = !(mirror || daemonset) && !drainable;
  if (blocks) f |= PF_BLOCKS_DRAIN;
  if (cnt > MAXSEL) f |= PF_SEL_OVERFLOW;

  int32_t nidx = -1;
  if (!node_name.empty()) {
    auto it = ctx.node_idx.find(std::string(node_name));
    if (it != ctx.node_idx.end()) {
      nidx = it->second;
      f |= PF_BOUND;
    }
  }
  out.pod_node.push_back(nidx);
  out.pod_flags.push_back(f);

  // pod LABEL pair hashes (host-only: PDB re-matching in the watch
  // cache) — sorted uint64 FNV pairs, first MAXPODLBL stored, RAW
  // count kept so >width rows are detectable (mirrors encode.py)
  const JValue* labels = get(meta, "labels");
  {
    std::vector<uint64_t> lh;
    if (labels && labels->kind == JValue::Obj) {
      for (auto& m : labels->members)
        if (m.second->kind == JValue::Str)
          lh.push_back(hash_label_pair(m.first, m.second->str));
    }
    std::sort(lh.begin(), lh.end());
    const int lcnt = (int)lh.size();
    for (int q = 0; q < MAXPODLBL; ++q)
      out.pod_label_pairs.push_back(q < lcnt ? (int64_t)lh[q] : 0);
    out.pod_label_count.push_back((int32_t)lcnt);
  }

  std::string ns(get_str(meta, "namespace"));
  if (ns.empty()) ns = "default";

  // PDB assignment: first SAME-NAMESPACE PDB whose matchLabels are
  // all present (v1 PDB semantics; mirrors encode.py/delta.py).
  // policy/v1: EMPTY matchLabels selects every pod in the namespace
  // (the inner loop is vacuously true).
  int32_t pdb = -1;
  if (!ctx.pdbs.empty()) {
    for (size_t d = 0; d < ctx.pdbs.size(); ++d) {
      if (ctx.pdbs[d].ns != ns) continue;
      bool all = true;
      for (auto& [k, v] : ctx.pdbs[d].labels) {
        const JValue* lv = labels ? get(labels, k) : nullptr;
        if (!lv || lv->kind != JValue::Str || lv->str != v) {
          all = false;
          break;
        }
      }
      if (all) {
        pdb = (int32_t)d;
        break;
      }
    }
  }
  out.pod_pdb.push_back(pdb);

  out.uids.emplace_back(get_str(meta, "uid"));
  out.ns.push_back(std::move(ns));
  out.names.emplace_back(get_str(meta, "name"));
}

// String-aware scan for the byte ranges of the items-array elements.
// Returns {items_found, [ (begin, end) per item ]}.
std::pair<bool, std::vector<std::pair<size_t, size_t>>> scan_items(
    std::string_view sv) {
  std::vector<std::pair<size_t, size_t>> spans;
  // locate "items" key at depth 1 with a full structural walk
  int depth = 0;
  bool in_str = false, esc = false;
  bool items_next = false;     // saw "items" key at depth 1, awaiting '['
  size_t i = 0, n = sv.size();
  size_t item_begin = 0;
  int items_depth = -1;        // depth of the items array once entered
  for (; i < n; ++i) {
    const char c = sv[i];
    if (in_str) {
      // memchr to the closing quote; a quote preceded by an odd number
      // of backslashes is escaped (strings are most of the bytes)
      if (esc) { esc = false; continue; }
      size_t j = i;
      while (true) {
        const void* q = memchr(sv.data() + j, '"', n - j);
        if (!q) { i = n; in_str = false; break; }
        size_t qi = (const char*)q - sv.data();
        size_t bs = 0;
        while (qi >= j + bs + 1 && sv[qi - bs - 1] == '\\') ++bs;
        if ((bs & 1) == 0) { i = qi; in_str = false; break; }
        j = qi + 1;
      }
      continue;
    }
    switch (c) {
      case '"': {
        if (depth == 1 && items_depth < 0) {
          // possible key start: match "items" followed by ':'
          if (n - i >= 7 && sv.compare(i, 7, "\"items\"") == 0) {
            size_t j = i + 7;
            while (j < n && (sv[j] == ' ' || sv[j] == '\t' || sv[j] == '\n' || sv[j] == '\r')) ++j;
            if (j < n && sv[j] == ':') items_next = true;
          }
        }
        in_str = true;
        break;
      }
      case '{':
        if (items_depth >= 0 && depth == items_depth) item_begin = i;
        ++depth;
        break;
      case '}':
        --depth;
        if (items_depth >= 0 && depth == items_depth) spans.emplace_back(item_begin, i + 1);
        break;
      case '[':
        if (items_next && items_depth < 0) {
          items_depth = depth + 1;
          items_next = false;
        }
        ++depth;
        break;
      case ']':
        --depth;
        if (items_depth >= 0 && depth == items_depth - 1) return {true, spans};
        break;
      default:
        break;
    }
  }
  return {items_depth >= 0, spans};
}

pybind11::dict encode_pods_json(pybind11::bytes body, pybind11::dict node_index,
                                pybind11::list pdb_selectors) {
  namespace py = pybind11;
  std::string_view sv = body;  // pybind keeps the bytes alive for the call

  Ctx ctx;
  for (auto item : node_index)
    ctx.node_idx[py::cast<std::string>(item.first)] = py::cast<int32_t>(item.second);
  parse_pdb_matchers(pdb_selectors, ctx);

  const bool dbg = getenv("MI355X_CODEC_DEBUG") != nullptr;
  auto t0 = std::chrono::steady_clock::now();
  std::vector<Columns> shards;
  {
    py::gil_scoped_release nogil;  // phases below are pure C++

    auto [found, spans] = scan_items(sv);
    if (!found) throw std::runtime_error("PodList has no items array");
    const size_t M = spans.size();

    unsigned hw = std::thread::hardware_concurrency();
    // up to 16 threads, >=1024 pods each: the 8-thread/4096-line cap
    // left most of the box idle on 10k-100k bursts
    size_t T = std::max<size_t>(1, std::min<size_t>({hw ? hw : 1, 16, (M + 1023) / 1024}));
    shards.resize(T);
    std::vector<std::thread> threads;
    std::atomic<bool> failed{false};
    std::string error_msg;
    std::mutex err_mu;
    for (size_t t = 0; t < T; ++t) {
      threads.emplace_back([&, t]() {
        const size_t lo = M * t / T, hi = M * (t + 1) / T;
        Columns& out = shards[t];
        out.reserve(hi - lo);
        try {
          Parser p(nullptr, 0);
          for (size_t k = lo; k < hi; ++k) {
            p.reset_to(sv.data() + spans[k].first, spans[k].second - spans[k].first);
            const JValue* pod = p.parse_value();
            encode_one(pod, ctx, out);
          }
        } catch (const std::exception& e) {
          std::lock_guard<std::mutex> g(err_mu);
          failed = true;
          error_msg = e.what();
        }
      });
    }
    for (auto& th : threads) th.join();
    if (failed) throw std::runtime_error("codec parse failed: " + error_msg);
  }
  auto t1 = std::chrono::steady_clock::now();

  // concatenate under the GIL into numpy arrays + python lists
  size_t P = 0;
  for (auto& s : shards) P += s.pod_flags.size();
  auto make = [&](auto getter, int width, auto tag) {
    using T = decltype(tag);
    py::array_t<T> arr(width == 1 ? std::vector<ssize_t>{(ssize_t)P}
                                  : std::vector<ssize_t>{(ssize_t)P, (ssize_t)width});
    T* dst = arr.mutable_data();
    for (auto& s : shards) {
      auto& v = getter(s);
      std::memcpy(dst, v.data(), v.size() * sizeof(T));
      dst += v.size();
    }
    return arr;
  };
  py::dict result;
  result["pod_req"] = make([](Columns& c) -> auto& { return c.pod_req; }, NRES, int32_t{});
  result["pod_flags"] = make([](Columns& c) -> auto& { return c.pod_flags; }, 1, int32_t{});
  result["pod_sel_hash"] = make([](Columns& c) -> auto& { return c.pod_sel_hash; }, 1, int64_t{});
  result["pod_sel_pairs"] = make([](Columns& c) -> auto& { return c.pod_sel_pairs; }, MAXSEL, int64_t{});
  result["pod_sel_count"] = make([](Columns& c) -> auto& { return c.pod_sel_count; }, 1, int32_t{});
  result["pod_node"] = make([](Columns& c) -> auto& { return c.pod_node; }, 1, int32_t{});
  result["pod_pdb"] = make([](Columns& c) -> auto& { return c.pod_pdb; }, 1, int32_t{});
  result["pod_label_pairs"] = make([](Columns& c) -> auto& { return c.pod_label_pairs; }, MAXPODLBL, int64_t{});
  result["pod_label_count"] = make([](Columns& c) -> auto& { return c.pod_label_count; }, 1, int32_t{});
  PyObject* uids = PyList_New((Py_ssize_t)P);
  PyObject* refs = PyList_New((Py_ssize_t)P);
  Py_ssize_t w = 0;
  for (auto& s : shards)
    for (size_t i = 0; i < s.uids.size(); ++i, ++w) {
      PyList_SET_ITEM(uids, w,
                      PyUnicode_FromStringAndSize(s.uids[i].data(), (Py_ssize_t)s.uids[i].size()));
      PyObject* tup = PyTuple_New(2);
      PyTuple_SET_ITEM(tup, 0,
                       PyUnicode_FromStringAndSize(s.ns[i].data(), (Py_ssize_t)s.ns[i].size()));
      PyTuple_SET_ITEM(tup, 1,
                       PyUnicode_FromStringAndSize(s.names[i].data(), (Py_ssize_t)s.names[i].size()));
      PyList_SET_ITEM(refs, w, tup);
    }
  result["pod_uids"] = py::reinterpret_steal<py::list>(uids);
  result["pod_refs"] = py::reinterpret_steal<py::list>(refs);
  if (dbg) {
    auto t2 = std::chrono::steady_clock::now();
    fprintf(stderr, "[codec] parse %.3fs gil-build %.3fs\n",
            std::chrono::duration<double>(t1 - t0).count(),
            std::chrono::duration<double>(t2 - t1).count());
  }
  return result;
}


// ---------------------------------------------------------------------------
// pack_into: memcpy a list of contiguous numpy arrays into a host
// staging buffer at given byte offsets — the 18-field python packing
// loop cost ~380 us/cycle at 10k pods; this is one call + 18 memcpys.
// ---------------------------------------------------------------------------
void pack_into(pybind11::buffer buf, const std::vector<int64_t>& offsets,
               pybind11::list arrays) {
  namespace py = pybind11;
  py::buffer_info dst = buf.request(true);
  char* base = static_cast<char*>(dst.ptr);
  const int64_t cap = dst.size * dst.itemsize;
  const size_t n = py::len(arrays);
  if (offsets.size() != n) throw std::runtime_error("offsets/arrays mismatch");
  for (size_t i = 0; i < n; ++i) {
    py::buffer src_b = py::cast<py::buffer>(arrays[i]);
    py::buffer_info src = src_b.request(false);
    int64_t nb = src.size * src.itemsize;
    if (offsets[i] + nb > cap) throw std::runtime_error("pack overflow");
    const bool strided2d =
        src.ndim == 2 && src.strides[1] == src.itemsize &&
        src.strides[0] != src.shape[1] * src.itemsize;
    if (strided2d) {
      // row-contiguous 2D view with a row gap (e.g. the narrowed
      // pod_sel_pairs[:, :W] slice): gather rows in one pass instead
      // of forcing an ascontiguousarray copy upstream
      const int64_t rows = src.shape[0];
      const size_t rowb = (size_t)(src.shape[1] * src.itemsize);
      const char* sp = static_cast<const char*>(src.ptr);
      char* dp = base + offsets[i];
      for (int64_t r = 0; r < rows; ++r)
        std::memcpy(dp + (size_t)r * rowb, sp + r * src.strides[0], rowb);
    } else {
      // contiguous (ascontiguousarray upstream)
      std::memcpy(base + offsets[i], src.ptr, (size_t)nb);
    }
  }
}


// ---------------------------------------------------------------------------
// decode_watch_events: JSONL watch stream ({"type":..,"object":{pod}})
// -> event types + the same pod columns as encode_pods_json, so the
// delta cache can batch-apply instead of per-event Python encoding.
// ---------------------------------------------------------------------------
pybind11::dict decode_watch_events(pybind11::bytes body,
                                   pybind11::dict node_index,
                                   pybind11::list pdb_selectors) {
  namespace py = pybind11;
  std::string_view sv = body;

  Ctx ctx;
  for (auto item : node_index)
    ctx.node_idx[py::cast<std::string>(item.first)] = py::cast<int32_t>(item.second);
  parse_pdb_matchers(pdb_selectors, ctx);

  struct EvShard {
    Columns cols;
    std::vector<int8_t> types;
  };
  std::vector<EvShard> shards;
  {
    py::gil_scoped_release nogil;
    // line spans (memchr scan), then thread-parallel per-line parse
    std::vector<std::pair<size_t, size_t>> lines;
    size_t pos = 0, n = sv.size();
    while (pos < n) {
      const void* q = memchr(sv.data() + pos, '\n', n - pos);
      size_t eol = q ? (const char*)q - sv.data() : n;
      if (eol > pos) lines.emplace_back(pos, eol);
      pos = eol + 1;
    }
    const size_t M = lines.size();
    unsigned hw = std::thread::hardware_concurrency();
    // up to 16 threads, >=512 events each (a 10k-event burst ran on
    // 5 threads under the old 2048-line shard floor)
    size_t T = std::max<size_t>(1, std::min<size_t>({hw ? hw : 1, 16, (M + 511) / 512}));
    shards.resize(T);
    std::vector<std::thread> threads;
    std::atomic<bool> failed{false};
    std::string error_msg;
    std::mutex err_mu;
    for (size_t t = 0; t < T; ++t) {
      threads.emplace_back([&, t]() {
        const size_t lo = M * t / T, hi = M * (t + 1) / T;
        EvShard& sh = shards[t];
        sh.cols.reserve(hi - lo);
        try {
          Parser p(nullptr, 0);
          for (size_t k = lo; k < hi; ++k) {
            p.reset_to(sv.data() + lines[k].first,
                       lines[k].second - lines[k].first);
            p.skip_ws();
            if (p.p_ >= p.end_) continue;
            const JValue* ev = p.parse_value();
            const std::string_view ty = get_str(ev, "type");
            const JValue* pod = get(ev, "object");
            if (pod && pod->kind == JValue::Obj) {
              encode_one(pod, ctx, sh.cols);
              sh.types.push_back(ty == "DELETED" ? 1
                                 : (ty == "ADDED" || ty == "MODIFIED") ? 0
                                                                       : -1);
            }
          }
        } catch (const std::exception& e) {
          std::lock_guard<std::mutex> g(err_mu);
          failed = true;
          error_msg = e.what();
        }
      });
    }
    for (auto& th : threads) th.join();
    if (failed) throw std::runtime_error("watch decode failed: " + error_msg);
  }
  Columns out;
  std::vector<int8_t> types;
  {
    size_t total = 0;
    for (auto& sh : shards) total += sh.types.size();
    out.reserve(total);
    types.reserve(total);
    for (auto& sh : shards) {
      auto app = [](auto& dst, auto& src) { dst.insert(dst.end(), src.begin(), src.end()); };
      app(out.pod_req, sh.cols.pod_req);
      app(out.pod_flags, sh.cols.pod_flags);
      app(out.pod_sel_hash, sh.cols.pod_sel_hash);
      app(out.pod_sel_pairs, sh.cols.pod_sel_pairs);
      app(out.pod_sel_count, sh.cols.pod_sel_count);
      app(out.pod_node, sh.cols.pod_node);
      app(out.pod_pdb, sh.cols.pod_pdb);
      app(out.pod_label_pairs, sh.cols.pod_label_pairs);
      app(out.pod_label_count, sh.cols.pod_label_count);
      for (auto& u : sh.cols.uids) out.uids.push_back(std::move(u));
      for (auto& x : sh.cols.ns) out.ns.push_back(std::move(x));
      for (auto& x : sh.cols.names) out.names.push_back(std::move(x));
      app(types, sh.types);
    }
  }

  const ssize_t P = (ssize_t)out.pod_flags.size();
  auto make = [&](auto& vec, int width, auto tag) {
    using T = decltype(tag);
    py::array_t<T> arr(width == 1 ? std::vector<ssize_t>{P}
                                  : std::vector<ssize_t>{P, (ssize_t)width});
    std::memcpy(arr.mutable_data(), vec.data(), vec.size() * sizeof(T));
    return arr;
  };
  py::dict result;
  result["types"] = make(types, 1, int8_t{});
  result["pod_req"] = make(out.pod_req, NRES, int32_t{});
  result["pod_flags"] = make(out.pod_flags, 1, int32_t{});
  result["pod_sel_hash"] = make(out.pod_sel_hash, 1, int64_t{});
  result["pod_sel_pairs"] = make(out.pod_sel_pairs, MAXSEL, int64_t{});
  result["pod_sel_count"] = make(out.pod_sel_count, 1, int32_t{});
  result["pod_node"] = make(out.pod_node, 1, int32_t{});
  result["pod_pdb"] = make(out.pod_pdb, 1, int32_t{});
  result["pod_label_pairs"] = make(out.pod_label_pairs, MAXPODLBL, int64_t{});
  result["pod_label_count"] = make(out.pod_label_count, 1, int32_t{});
  PyObject* uids = PyList_New(P);
  PyObject* refs = PyList_New(P);
  for (ssize_t i = 0; i < P; ++i) {
    PyList_SET_ITEM(uids, i,
                    PyUnicode_FromStringAndSize(out.uids[i].data(), (Py_ssize_t)out.uids[i].size()));
    PyObject* tup = PyTuple_New(2);
    PyTuple_SET_ITEM(tup, 0,
                     PyUnicode_FromStringAndSize(out.ns[i].data(), (Py_ssize_t)out.ns[i].size()));
    PyTuple_SET_ITEM(tup, 1,
                     PyUnicode_FromStringAndSize(out.names[i].data(), (Py_ssize_t)out.names[i].size()));
    PyList_SET_ITEM(refs, i, tup);
  }
  result["pod_uids"] = py::reinterpret_steal<py::list>(uids);
  result["pod_refs"] = py::reinterpret_steal<py::list>(refs);
  return result;
}

// ---------------------------------------------------------------------------
// Batch row resolution for DeltaSnapshotCache.apply_batch: the
// per-event dict/list bookkeeping in C under the GIL (the Python loop
// bounded watch ingest at ~580k events/s after the threaded JSONL
// decode). Mutates uid_row / free_rows / pod_uids / pod_refs exactly
// as the Python reference loop (kept in delta.py as the fallback) and
// returns (rows, sel, cleared, new_size): `rows[j] = target row of
// upsert event sel[j]` for the caller's vectorized column writes, and
// `cleared` = rows tombstoned by DELETED events (caller zeroes the
// columns). The caller must pre-grow so that capacity can never be
// exhausted mid-batch.
py::tuple resolve_batch_rows(
    py::array_t<int8_t, py::array::c_style | py::array::forcecast> types_arr,
    py::list uids, py::list refs, py::dict uid_row, py::list free_rows,
    int64_t size, int64_t cap, py::list pod_uids, py::list pod_refs) {
  const int8_t* t = types_arr.data();
  const py::ssize_t n = types_arr.shape(0);
  PyObject* p_uidrow = uid_row.ptr();
  PyObject* p_free = free_rows.ptr();
  PyObject* p_uids_out = pod_uids.ptr();
  PyObject* p_refs_out = pod_refs.ptr();
  std::vector<int64_t> rows, sel, cleared;
  rows.reserve((size_t)n);
  sel.reserve((size_t)n);
  static PyObject* empty_str = PyUnicode_InternFromString("");
  for (py::ssize_t i = 0; i < n; ++i) {
    const int ty = t[i];
    PyObject* u = PyList_GET_ITEM(uids.ptr(), i);  // borrowed
    if (ty == 1) {  // DELETED
      PyObject* r = PyDict_GetItemWithError(p_uidrow, u);
      if (!r) {
        if (PyErr_Occurred()) throw py::error_already_set();
        continue;
      }
      const long long row = PyLong_AsLongLong(r);
      if (PyDict_DelItem(p_uidrow, u) < 0) throw py::error_already_set();
      {
        py::int_ rv(row);
        if (PyList_Append(p_free, rv.ptr()) < 0) throw py::error_already_set();
      }
      Py_INCREF(empty_str);
      PyList_SetItem(p_uids_out, (Py_ssize_t)row, empty_str);
      PyObject* tup = PyTuple_New(2);
      Py_INCREF(empty_str);
      Py_INCREF(empty_str);
      PyTuple_SET_ITEM(tup, 0, empty_str);
      PyTuple_SET_ITEM(tup, 1, empty_str);
      PyList_SetItem(p_refs_out, (Py_ssize_t)row, tup);
      cleared.push_back(row);
      continue;
    }
    if (ty != 0) continue;  // BOOKMARK / unknown kinds
    PyObject* r = PyDict_GetItemWithError(p_uidrow, u);
    if (!r && PyErr_Occurred()) throw py::error_already_set();
    long long row;
    if (r) {
      row = PyLong_AsLongLong(r);
    } else {
      const Py_ssize_t fn = PyList_GET_SIZE(p_free);
      if (fn > 0) {
        row = PyLong_AsLongLong(PyList_GET_ITEM(p_free, fn - 1));
        if (PySequence_DelItem(p_free, fn - 1) < 0) throw py::error_already_set();
      } else {
        if (size >= cap)
          throw std::runtime_error(
              "resolve_batch_rows: capacity exhausted (caller must pre-grow)");
        row = size++;
      }
      py::int_ rv(row);
      if (PyDict_SetItem(p_uidrow, u, rv.ptr()) < 0) throw py::error_already_set();
    }
    rows.push_back(row);
    sel.push_back(i);
    Py_INCREF(u);
    PyList_SetItem(p_uids_out, (Py_ssize_t)row, u);
    PyObject* rf = PyList_GET_ITEM(refs.ptr(), i);
    Py_INCREF(rf);
    PyList_SetItem(p_refs_out, (Py_ssize_t)row, rf);
  }
  auto mk = [](std::vector<int64_t>& v) {
    py::array_t<int64_t> a((py::ssize_t)v.size());
    if (!v.empty()) std::memcpy(a.mutable_data(), v.data(), v.size() * sizeof(int64_t));
    return a;
  };
  return py::make_tuple(mk(rows), mk(sel), mk(cleared), (int64_t)size);
}
