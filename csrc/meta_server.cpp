// Native metadata RPC frontend: GIL-free epoll threads own the master's
// client sockets and serve the hot read-only metadata ops (FileStatus=7,
// ListStatus=8, Exists=9, keepalive=1) straight from a C++ mirror of the
// inode tree; every other frame is forwarded to the Python handler and its
// reply is written back through the same connection.
//
// The MI355X analog of the reference's tokio/prost RPC server hot path
// (crates/core/rpc/src/server/rpc_server.rs:27-232): metadata QPS is
// request-parse + tree-walk + response-encode bound, none of which needs
// the interpreter.  Python pushes pre-encoded msgpack field blobs (one per
// inode, all FileStatus fields except the lookup-dependent "path") on
// every mutation apply, so a served status reply is a header splice, not a
// re-serialization.
//
// Consistency contract with Python (curvine_amd/master/native_meta.py):
//  * mirror updates happen synchronously inside FsDir._apply_* (the same
//    choke point live execution and journal replay share), before the
//    mutating RPC's reply is sent — read-your-writes holds exactly as it
//    does for the Python-served path;
//  * the serving flag is flipped under the tree's unique lock, so a
//    leadership step-down (or mirror rebuild) never races in-flight
//    serves: set_serving(false) returns only after the last reader left.

#include <arpa/inet.h>
#include <fcntl.h>
#include <sys/eventfd.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <map>
#include <memory>
#include <mutex>
#include <shared_mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

// ---------------------------------------------------------------- wire

static constexpr size_t kMetaProto = 22;
static constexpr uint32_t kMetaMaxLen = 16u << 20;
static constexpr uint8_t kCodeKeepalive = 1;
static constexpr uint8_t kCodeOpenFile = 5;
static constexpr uint8_t kCodeFileStatus = 7;
static constexpr uint8_t kCodeListStatus = 8;
static constexpr uint8_t kCodeExists = 9;
static constexpr uint8_t kCodeGetBlockLocations = 13;
static constexpr uint8_t kErrFileNotFound = 3;  // errors.ErrorCode
static constexpr uint8_t kErrIsDirectory = 7;
static constexpr int64_t kRootId = 1;

static inline uint32_t rd_u32be(const uint8_t* p) {
  return (uint32_t(p[0]) << 24) | (uint32_t(p[1]) << 16) |
         (uint32_t(p[2]) << 8) | uint32_t(p[3]);
}
static inline uint64_t rd_u64be(const uint8_t* p) {
  return (uint64_t(rd_u32be(p)) << 32) | rd_u32be(p + 4);
}
static inline void wr_u16be(std::string& o, uint16_t v) {
  o.push_back(char(v >> 8));
  o.push_back(char(v));
}
static inline void wr_u32be(std::string& o, uint32_t v) {
  o.push_back(char(v >> 24));
  o.push_back(char(v >> 16));
  o.push_back(char(v >> 8));
  o.push_back(char(v));
}
static inline void wr_u64be(std::string& o, uint64_t v) {
  wr_u32be(o, uint32_t(v >> 32));
  wr_u32be(o, uint32_t(v));
}

static void mp_str(std::string& o, const char* s, size_t n) {
  if (n < 32) {
    o.push_back(char(0xa0 | n));
  } else if (n < 256) {
    o.push_back(char(0xd9));
    o.push_back(char(n));
  } else if (n < 65536) {
    o.push_back(char(0xda));
    wr_u16be(o, uint16_t(n));
  } else {
    o.push_back(char(0xdb));
    wr_u32be(o, uint32_t(n));
  }
  o.append(s, n);
}
static inline void mp_str(std::string& o, const std::string& s) {
  mp_str(o, s.data(), s.size());
}
static void mp_uint(std::string& o, uint64_t v) {
  if (v < 128) {
    o.push_back(char(v));
  } else if (v < 256) {
    o.push_back(char(0xcc));
    o.push_back(char(v));
  } else if (v < 65536) {
    o.push_back(char(0xcd));
    wr_u16be(o, uint16_t(v));
  } else if (v <= 0xffffffffu) {
    o.push_back(char(0xce));
    wr_u32be(o, uint32_t(v));
  } else {
    o.push_back(char(0xcf));
    wr_u64be(o, v);
  }
}

// minimal msgpack skip/scan for request headers ({"path": str, "cid": ...})
static bool mp_skip(const uint8_t*& p, const uint8_t* end);

static bool mp_skip_n(const uint8_t*& p, const uint8_t* end, size_t n) {
  for (size_t i = 0; i < n; i++)
    if (!mp_skip(p, end)) return false;
  return true;
}

static bool mp_skip(const uint8_t*& p, const uint8_t* end) {
  if (p >= end) return false;
  uint8_t b = *p++;
  if (b <= 0x7f || b >= 0xe0) return true;           // fixint
  if (b >= 0x80 && b <= 0x8f) return mp_skip_n(p, end, size_t(b & 0xf) * 2);
  if (b >= 0x90 && b <= 0x9f) return mp_skip_n(p, end, b & 0xf);
  if (b >= 0xa0 && b <= 0xbf) {                      // fixstr
    size_t n = b & 0x1f;
    if (end - p < (ptrdiff_t)n) return false;
    p += n;
    return true;
  }
  auto need = [&](size_t n) -> bool {
    if (end - p < (ptrdiff_t)n) return false;
    p += n;
    return true;
  };
  switch (b) {
    case 0xc0: case 0xc2: case 0xc3: return true;    // nil/bool
    case 0xcc: case 0xd0: return need(1);
    case 0xcd: case 0xd1: return need(2);
    case 0xce: case 0xd2: case 0xca: return need(4);
    case 0xcf: case 0xd3: case 0xcb: return need(8);
    case 0xd9: case 0xc4: {                          // str8/bin8
      if (p >= end) return false;
      size_t n = *p++;
      return need(n);
    }
    case 0xda: case 0xc5: {
      if (end - p < 2) return false;
      size_t n = (size_t(p[0]) << 8) | p[1];
      p += 2;
      return need(n);
    }
    case 0xdb: case 0xc6: {
      if (end - p < 4) return false;
      size_t n = rd_u32be(p);
      p += 4;
      return need(n);
    }
    case 0xdc: {
      if (end - p < 2) return false;
      size_t n = (size_t(p[0]) << 8) | p[1];
      p += 2;
      return mp_skip_n(p, end, n);
    }
    case 0xdd: {
      if (end - p < 4) return false;
      size_t n = rd_u32be(p);
      p += 4;
      return mp_skip_n(p, end, n);
    }
    case 0xde: {
      if (end - p < 2) return false;
      size_t n = (size_t(p[0]) << 8) | p[1];
      p += 2;
      return mp_skip_n(p, end, n * 2);
    }
    case 0xdf: {
      if (end - p < 4) return false;
      size_t n = rd_u32be(p);
      p += 4;
      return mp_skip_n(p, end, n * 2);
    }
    default:
      return false;  // ext & friends: unsupported -> caller forwards
  }
}

static bool mp_read_str(const uint8_t*& p, const uint8_t* end,
                        const char** s, size_t* n) {
  if (p >= end) return false;
  uint8_t b = *p++;
  size_t len;
  if (b >= 0xa0 && b <= 0xbf) {
    len = b & 0x1f;
  } else if (b == 0xd9) {
    if (p >= end) return false;
    len = *p++;
  } else if (b == 0xda) {
    if (end - p < 2) return false;
    len = (size_t(p[0]) << 8) | p[1];
    p += 2;
  } else if (b == 0xdb) {
    if (end - p < 4) return false;
    len = rd_u32be(p);
    p += 4;
  } else {
    return false;
  }
  if (end - p < (ptrdiff_t)len) return false;
  *s = (const char*)p;
  *n = len;
  p += len;
  return true;
}

// scan the top-level request header map for "path"
static bool mp_find_path(const uint8_t* p, const uint8_t* end,
                         std::string* path) {
  if (p >= end) return false;
  uint8_t b = *p++;
  size_t pairs;
  if (b >= 0x80 && b <= 0x8f) {
    pairs = b & 0xf;
  } else if (b == 0xde) {
    if (end - p < 2) return false;
    pairs = (size_t(p[0]) << 8) | p[1];
    p += 2;
  } else if (b == 0xdf) {
    if (end - p < 4) return false;
    pairs = rd_u32be(p);
    p += 4;
  } else {
    return false;
  }
  for (size_t i = 0; i < pairs; i++) {
    const char* ks;
    size_t kn;
    if (!mp_read_str(p, end, &ks, &kn)) return false;
    if (kn == 4 && memcmp(ks, "path", 4) == 0) {
      const char* vs;
      size_t vn;
      if (!mp_read_str(p, end, &vs, &vn)) return false;
      path->assign(vs, vn);
      return true;
    }
    if (!mp_skip(p, end)) return false;
  }
  return false;
}

// ---------------------------------------------------------------- state

struct MetaNode {
  bool is_dir = false;
  uint32_t npairs = 0;
  int64_t mtime = 0;   // kept OUT of blob: parent-mtime touches are O(1)
  std::string blob;                      // msgpack pairs sans path/mtime
  std::map<std::string, int64_t> children;  // sorted == Python sorted()
  std::vector<std::pair<int64_t, int64_t>> blocks;  // (block_id, length)
};

struct LocEnt {
  int64_t wid;
  uint8_t ord;        // TIER_ORDER rank (stable-sort key, hottest first)
  std::string tier;
};

struct WorkerRec {
  std::string blob;   // msgpack map of WorkerAddress.to_dict()
  bool lost = false;
};

struct MetaConn {
  int fd = -1;
  uint64_t id = 0;
  std::string rbuf;
  std::mutex wmu;
  std::atomic<bool> dead{false};
};

struct MetaServer {
  int listen_fd = -1, epfd = -1;
  std::atomic<bool> stopping{false};
  bool serving = false;  // guarded by tree_mu
  std::shared_mutex tree_mu;
  std::unordered_map<int64_t, MetaNode> nodes;
  std::unordered_map<int64_t, WorkerRec> workers;        // guarded by tree_mu
  std::unordered_map<int64_t, std::vector<LocEnt>> block_locs;
  std::mutex acc_mu;   // open() access counters, drained by Python
  std::unordered_map<int64_t, uint64_t> access_counts;
  std::mutex conns_mu;
  std::unordered_map<uint64_t, std::shared_ptr<MetaConn>> conns;
  std::atomic<uint64_t> next_conn{1};
  std::vector<std::thread> threads;
  std::mutex fq_mu;
  std::condition_variable fq_cv;
  std::deque<std::pair<uint64_t, std::string>> fq;
  int fq_efd = -1;   // eventfd: wakes the Python loop's add_reader
  std::atomic<uint64_t> served_status{0}, served_list{0}, served_exists{0},
      served_open{0}, served_ping{0}, served_notfound{0}, forwarded{0},
      conns_total{0};
};

static std::mutex g_meta_mu;
// leaked on purpose: destroying a MetaServer with live threads at
// interpreter teardown would std::terminate; meta_stop() is the real
// cleanup path
static auto& g_meta =
    *new std::unordered_map<int64_t, std::unique_ptr<MetaServer>>();
static int64_t g_meta_next = 1;

static MetaServer* meta_get(int64_t sid) {
  std::lock_guard<std::mutex> g(g_meta_mu);
  auto it = g_meta.find(sid);
  if (it == g_meta.end()) throw std::runtime_error("bad meta server id");
  return it->second.get();
}

// ---------------------------------------------------------------- io

static bool meta_write_all(MetaConn* c, const char* p, size_t n) {
  std::lock_guard<std::mutex> g(c->wmu);
  if (c->dead.load()) return false;
  while (n) {
    ssize_t w = send(c->fd, p, n, MSG_NOSIGNAL);
    if (w > 0) {
      p += w;
      n -= size_t(w);
      continue;
    }
    if (w < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
      struct pollfd pf = {c->fd, POLLOUT, 0};
      if (poll(&pf, 1, 10000) <= 0) return false;
      continue;
    }
    if (w < 0 && errno == EINTR) continue;
    return false;
  }
  return true;
}

static void meta_close_conn(MetaServer* S, const std::shared_ptr<MetaConn>& c) {
  bool was = c->dead.exchange(true);
  if (was) return;
  epoll_ctl(S->epfd, EPOLL_CTL_DEL, c->fd, nullptr);
  close(c->fd);
  {
    std::lock_guard<std::mutex> g(S->conns_mu);
    S->conns.erase(c->id);
  }
  {
    // empty-frame sentinel: tells Python to reap this conn's drain task
    std::lock_guard<std::mutex> g(S->fq_mu);
    S->fq.emplace_back(c->id, std::string());
  }
  S->fq_cv.notify_one();
  if (S->fq_efd >= 0) {
    uint64_t one = 1;
    ssize_t r = write(S->fq_efd, &one, 8);
    (void)r;
  }
}

static std::string meta_proto(uint32_t hlen, uint32_t dlen, uint8_t code,
                              uint8_t status, uint64_t req_id, uint32_t seq) {
  std::string o;
  o.reserve(kMetaProto + hlen);
  wr_u32be(o, hlen);
  wr_u32be(o, dlen);
  o.push_back(char(code));
  o.push_back(char(status));
  wr_u64be(o, req_id);
  wr_u32be(o, seq);
  return o;
}

static void meta_reply(MetaServer* S, MetaConn* c, uint8_t code,
                       uint8_t req_status, uint64_t req_id, uint32_t seq,
                       const std::string& hdr, bool error = false) {
  uint8_t status = uint8_t(((error ? 5 : 3) << 4) | (req_status & 0xF));
  std::string out = meta_proto(uint32_t(hdr.size()), 0, code, status,
                               req_id, seq);
  out += hdr;
  if (!meta_write_all(c, out.data(), out.size())) c->dead.store(true);
}

static void meta_reply_fserr(MetaServer* S, MetaConn* c, uint8_t code,
                             uint8_t req_status, uint64_t req_id,
                             uint32_t seq, uint8_t err_code,
                             const std::string& msg) {
  std::string h;
  h.push_back(char(0x82));
  mp_str(h, "error_code", 10);
  mp_uint(h, err_code);
  mp_str(h, "error_msg", 9);
  mp_str(h, msg);
  S->served_notfound.fetch_add(1, std::memory_order_relaxed);
  meta_reply(S, c, code, req_status, req_id, seq, h, /*error=*/true);
}

// ---------------------------------------------------------------- tree

// 0 = ok, 1 = not found, 2 = forward to Python (needs norm/validation)
static int meta_resolve(MetaServer* S, const std::string& path,
                        std::string* norm, int64_t* out) {
  if (path.empty() || path[0] != '/') return 2;
  int64_t cur = kRootId;
  auto root = S->nodes.find(cur);
  if (root == S->nodes.end()) return 2;  // mirror not primed
  norm->clear();
  const MetaNode* nd = &root->second;
  size_t i = 1;
  while (i < path.size()) {
    size_t j = path.find('/', i);
    if (j == std::string::npos) j = path.size();
    size_t n = j - i;
    if (n == 0 || (n == 1 && path[i] == '.')) {  // "//" and "/./"
      i = j + 1;
      continue;
    }
    if (n == 2 && path[i] == '.' && path[i + 1] == '.') return 2;  // ".."
    if (!nd->is_dir) return 1;
    auto ch = nd->children.find(path.substr(i, n));
    if (ch == nd->children.end()) return 1;
    cur = ch->second;
    auto it = S->nodes.find(cur);
    if (it == S->nodes.end()) return 2;  // mirror edge without node: bail
    nd = &it->second;
    norm->push_back('/');
    norm->append(path, i, n);
    i = j + 1;
  }
  if (norm->empty()) *norm = "/";
  *out = cur;
  return 0;
}

static void meta_append_status(std::string& o, const MetaNode& nd,
                               const std::string& path) {
  o.push_back(char(0xde));
  wr_u16be(o, uint16_t(nd.npairs + 2));
  mp_str(o, "path", 4);
  mp_str(o, path);
  mp_str(o, "mtime_ms", 8);
  mp_uint(o, uint64_t(nd.mtime));
  o += nd.blob;
}

// returns false -> caller forwards the frame to Python
static bool meta_serve(MetaServer* S, MetaConn* c, const uint8_t* frame,
                       uint32_t hlen, uint8_t code, uint8_t status,
                       uint64_t req_id, uint32_t seq) {
  uint8_t req_status = status & 0xF;
  if (req_status != 0) return false;  // streaming: not ours
  std::string path;
  if (!mp_find_path(frame + kMetaProto, frame + kMetaProto + hlen, &path))
    return false;
  std::shared_lock<std::shared_mutex> lk(S->tree_mu);
  if (!S->serving) return false;
  std::string norm;
  int64_t id = 0;
  int rc = meta_resolve(S, path, &norm, &id);
  if (rc == 2) return false;
  if (rc == 1) {
    if (code == kCodeExists) {
      std::string h;
      h.push_back(char(0x81));
      mp_str(h, "exists", 6);
      h.push_back(char(0xc2));
      S->served_exists.fetch_add(1, std::memory_order_relaxed);
      meta_reply(S, c, code, req_status, req_id, seq, h);
      return true;
    }
    lk.unlock();
    meta_reply_fserr(S, c, code, req_status, req_id, seq, kErrFileNotFound,
                     path);
    return true;
  }
  const MetaNode& nd = S->nodes.find(id)->second;
  std::string h;
  if (code == kCodeOpenFile || code == kCodeGetBlockLocations) {
    if (nd.is_dir) {
      lk.unlock();
      meta_reply_fserr(S, c, code, req_status, req_id, seq, kErrIsDirectory,
                       path);
      return true;
    }
    h.push_back(char(0x81));
    mp_str(h, "file_blocks", 11);
    h.push_back(char(0x82));
    mp_str(h, "status", 6);
    meta_append_status(h, nd, norm);
    mp_str(h, "blocks", 6);
    size_t nb = nd.blocks.size();
    if (nb < 16) {
      h.push_back(char(0x90 | nb));
    } else if (nb < 65536) {
      h.push_back(char(0xdc));
      wr_u16be(h, uint16_t(nb));
    } else {
      h.push_back(char(0xdd));
      wr_u32be(h, uint32_t(nb));
    }
    uint64_t off = 0;
    std::vector<const LocEnt*> locs;
    for (const auto& blk : nd.blocks) {
      // live locations, hottest tier first (locations_of analog; stable
      // sort keeps heartbeat insertion order within a tier)
      locs.clear();
      auto bl = S->block_locs.find(blk.first);
      if (bl != S->block_locs.end()) {
        for (const auto& e : bl->second) {
          auto w = S->workers.find(e.wid);
          if (w != S->workers.end() && !w->second.lost) locs.push_back(&e);
        }
        std::stable_sort(locs.begin(), locs.end(),
                         [](const LocEnt* a, const LocEnt* b) {
                           return a->ord < b->ord;
                         });
      }
      h.push_back(char(0x84));
      mp_str(h, "block", 5);
      h.push_back(char(0x83));
      mp_str(h, "block_id", 8);
      mp_uint(h, uint64_t(blk.first));
      mp_str(h, "length", 6);
      mp_uint(h, uint64_t(blk.second));
      mp_str(h, "state", 5);
      h.push_back(char(0x01));              // BlockState.FINALIZED
      mp_str(h, "offset", 6);
      mp_uint(h, off);
      mp_str(h, "locations", 9);
      size_t nl = locs.size();
      if (nl < 16) {
        h.push_back(char(0x90 | nl));
      } else {
        h.push_back(char(0xdc));
        wr_u16be(h, uint16_t(nl));
      }
      for (const LocEnt* e : locs) h += S->workers.find(e->wid)->second.blob;
      mp_str(h, "tiers", 5);
      if (nl < 16) {
        h.push_back(char(0x90 | nl));
      } else {
        h.push_back(char(0xdc));
        wr_u16be(h, uint16_t(nl));
      }
      for (const LocEnt* e : locs) mp_str(h, e->tier);
      off += uint64_t(blk.second);
    }
    S->served_open.fetch_add(1, std::memory_order_relaxed);
    lk.unlock();
    {
      std::lock_guard<std::mutex> g(S->acc_mu);
      S->access_counts[id]++;
    }
    meta_reply(S, c, code, req_status, req_id, seq, h);
    return true;
  }
  if (code == kCodeExists) {
    h.push_back(char(0x81));
    mp_str(h, "exists", 6);
    h.push_back(char(0xc3));
    S->served_exists.fetch_add(1, std::memory_order_relaxed);
  } else if (code == kCodeFileStatus) {
    h.push_back(char(0x81));
    mp_str(h, "status", 6);
    meta_append_status(h, nd, norm);
    S->served_status.fetch_add(1, std::memory_order_relaxed);
  } else {  // ListStatus
    h.push_back(char(0x81));
    mp_str(h, "statuses", 8);
    if (!nd.is_dir) {
      h.push_back(char(0x91));
      meta_append_status(h, nd, norm);
    } else {
      size_t n = nd.children.size();
      if (n < 65536) {
        h.push_back(char(0xdc));
        wr_u16be(h, uint16_t(n));
      } else {
        h.push_back(char(0xdd));
        wr_u32be(h, uint32_t(n));
      }
      std::string base = (norm == "/") ? "" : norm;
      std::string cpath;
      for (const auto& kv : nd.children) {
        auto it = S->nodes.find(kv.second);
        if (it == S->nodes.end()) return false;  // torn mirror: forward
        cpath = base;
        cpath.push_back('/');
        cpath += kv.first;
        meta_append_status(h, it->second, cpath);
      }
    }
    S->served_list.fetch_add(1, std::memory_order_relaxed);
  }
  lk.unlock();
  meta_reply(S, c, code, req_status, req_id, seq, h);
  return true;
}

static void meta_handle_frame(MetaServer* S, const std::shared_ptr<MetaConn>& c,
                              const uint8_t* frame, uint32_t hlen,
                              uint32_t dlen) {
  uint8_t code = frame[8], status = frame[9];
  uint64_t req_id = rd_u64be(frame + 10);
  uint32_t seq = rd_u32be(frame + 18);
  if (code == kCodeKeepalive && hlen == 0 && dlen == 0) {
    std::string out =
        meta_proto(0, 0, code, uint8_t((3 << 4) | (status & 0xF)), req_id, seq);
    S->served_ping.fetch_add(1, std::memory_order_relaxed);
    if (!meta_write_all(c.get(), out.data(), out.size())) c->dead.store(true);
    return;
  }
  if ((code == kCodeFileStatus || code == kCodeListStatus ||
       code == kCodeExists || code == kCodeOpenFile ||
       code == kCodeGetBlockLocations) &&
      dlen == 0) {
    if (meta_serve(S, c.get(), frame, hlen, code, status, req_id, seq)) return;
  }
  S->forwarded.fetch_add(1, std::memory_order_relaxed);
  bool was_empty;
  {
    std::lock_guard<std::mutex> g(S->fq_mu);
    was_empty = S->fq.empty();
    S->fq.emplace_back(c->id,
                       std::string((const char*)frame,
                                   kMetaProto + hlen + dlen));
  }
  S->fq_cv.notify_one();
  if (was_empty && S->fq_efd >= 0) {
    uint64_t one = 1;
    ssize_t r = write(S->fq_efd, &one, 8);
    (void)r;
  }
}

// ---------------------------------------------------------------- loop

static void meta_readable(MetaServer* S, const std::shared_ptr<MetaConn>& c) {
  char tmp[128 << 10];
  for (;;) {
    ssize_t n = recv(c->fd, tmp, sizeof tmp, 0);
    if (n > 0) {
      c->rbuf.append(tmp, size_t(n));
      if (size_t(n) < sizeof tmp) break;
      continue;
    }
    if (n == 0) {
      meta_close_conn(S, c);
      return;
    }
    if (errno == EINTR) continue;
    if (errno == EAGAIN || errno == EWOULDBLOCK) break;
    meta_close_conn(S, c);
    return;
  }
  size_t off = 0;
  const uint8_t* buf = (const uint8_t*)c->rbuf.data();
  size_t avail = c->rbuf.size();
  while (avail - off >= kMetaProto) {
    uint32_t hlen = rd_u32be(buf + off);
    uint32_t dlen = rd_u32be(buf + off + 4);
    if (hlen > kMetaMaxLen || dlen > kMetaMaxLen) {
      meta_close_conn(S, c);
      return;
    }
    size_t total = kMetaProto + hlen + dlen;
    if (avail - off < total) break;
    meta_handle_frame(S, c, buf + off, hlen, dlen);
    off += total;
  }
  if (off) c->rbuf.erase(0, off);
  if (c->dead.load()) {
    meta_close_conn(S, c);
    return;
  }
  struct epoll_event ev;
  ev.events = EPOLLIN | EPOLLRDHUP | EPOLLONESHOT;
  ev.data.u64 = c->id;
  if (epoll_ctl(S->epfd, EPOLL_CTL_MOD, c->fd, &ev) != 0)
    meta_close_conn(S, c);
}

static void meta_accept(MetaServer* S) {
  for (;;) {
    int fd = accept4(S->listen_fd, nullptr, nullptr, SOCK_NONBLOCK);
    if (fd < 0) break;
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
    auto c = std::make_shared<MetaConn>();
    c->fd = fd;
    c->id = S->next_conn.fetch_add(1);
    {
      std::lock_guard<std::mutex> g(S->conns_mu);
      S->conns[c->id] = c;
    }
    S->conns_total.fetch_add(1, std::memory_order_relaxed);
    struct epoll_event ev;
    ev.events = EPOLLIN | EPOLLRDHUP | EPOLLONESHOT;
    ev.data.u64 = c->id;
    if (epoll_ctl(S->epfd, EPOLL_CTL_ADD, fd, &ev) != 0) meta_close_conn(S, c);
  }
  struct epoll_event ev;
  ev.events = EPOLLIN | EPOLLONESHOT;
  ev.data.u64 = 0;
  epoll_ctl(S->epfd, EPOLL_CTL_MOD, S->listen_fd, &ev);
}

static void meta_thread(MetaServer* S) {
  while (!S->stopping.load()) {
    struct epoll_event ev;
    int n = epoll_wait(S->epfd, &ev, 1, 500);
    if (n <= 0) continue;
    if (ev.data.u64 == 0) {
      meta_accept(S);
      continue;
    }
    std::shared_ptr<MetaConn> c;
    {
      std::lock_guard<std::mutex> g(S->conns_mu);
      auto it = S->conns.find(ev.data.u64);
      if (it != S->conns.end()) c = it->second;
    }
    if (!c) continue;
    if (ev.events & (EPOLLHUP | EPOLLERR)) {
      meta_close_conn(S, c);
      continue;
    }
    meta_readable(S, c);
  }
}

// ---------------------------------------------------------------- api

static int64_t meta_create(int listen_fd, int nthreads) {
  auto S = std::make_unique<MetaServer>();
  S->listen_fd = listen_fd;
  S->fq_efd = eventfd(0, EFD_NONBLOCK | EFD_CLOEXEC);
  int fl = fcntl(listen_fd, F_GETFL, 0);
  fcntl(listen_fd, F_SETFL, fl | O_NONBLOCK);
  S->epfd = epoll_create1(EPOLL_CLOEXEC);
  if (S->epfd < 0) throw std::runtime_error("epoll_create1 failed");
  struct epoll_event ev;
  ev.events = EPOLLIN | EPOLLONESHOT;
  ev.data.u64 = 0;
  if (epoll_ctl(S->epfd, EPOLL_CTL_ADD, listen_fd, &ev) != 0)
    throw std::runtime_error("epoll_ctl(listen) failed");
  MetaServer* raw = S.get();
  for (int i = 0; i < nthreads; i++)
    S->threads.emplace_back(meta_thread, raw);
  std::lock_guard<std::mutex> g(g_meta_mu);
  int64_t sid = g_meta_next++;
  g_meta[sid] = std::move(S);
  return sid;
}

static void meta_stop_srv(int64_t sid) {
  std::unique_ptr<MetaServer> S;
  {
    std::lock_guard<std::mutex> g(g_meta_mu);
    auto it = g_meta.find(sid);
    if (it == g_meta.end()) return;
    S = std::move(it->second);
    g_meta.erase(it);
  }
  S->stopping.store(true);
  S->fq_cv.notify_all();
  {
    py::gil_scoped_release rel;
    for (auto& t : S->threads) t.join();
  }
  close(S->listen_fd);
  close(S->epfd);
  if (S->fq_efd >= 0) close(S->fq_efd);
  for (auto& kv : S->conns) close(kv.second->fd);
}

static void meta_set_serving(int64_t sid, bool on) {
  MetaServer* S = meta_get(sid);
  py::gil_scoped_release rel;
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  S->serving = on;
}

static void meta_upsert(int64_t sid, int64_t id, bool is_dir,
                        py::bytes blob, uint32_t npairs, py::bytes blocks,
                        int64_t mtime) {
  MetaServer* S = meta_get(sid);
  std::string b = blob;
  std::string bb = blocks;  // packed little-endian (block_id, length) i64 pairs
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  MetaNode& nd = S->nodes[id];
  nd.is_dir = is_dir;
  nd.npairs = npairs;
  nd.mtime = mtime;
  nd.blob = std::move(b);
  nd.blocks.clear();
  const int64_t* p = (const int64_t*)bb.data();
  size_t n = bb.size() / 16;
  nd.blocks.reserve(n);
  for (size_t i = 0; i < n; i++) nd.blocks.emplace_back(p[2 * i], p[2 * i + 1]);
}

static void mp_kv_uint(std::string& o, const char* k, size_t kn, uint64_t v) {
  mp_str(o, k, kn);
  mp_uint(o, v);
}

// fast-path upsert: the FileStatus blob is packed HERE from positional
// args (no Python dict + packb per mutation — the mutation-QPS hot path)
static void meta_upsert_node(int64_t sid, int64_t id, bool is_dir,
                             const std::string& name, int64_t file_type,
                             int64_t length, bool is_complete,
                             int64_t block_size, int64_t replicas,
                             const std::string& storage_tier,
                             int64_t mtime_ms, int64_t atime_ms, int64_t mode,
                             int64_t uid, int64_t gid, int64_t ttl_ms,
                             const std::string& ttl_action,
                             const std::string& symlink_target,
                             int64_t nlink, py::bytes blocks,
                             py::object xattrs_blob) {
  MetaServer* S = meta_get(sid);
  std::string bb = blocks;
  std::string blob;
  blob.reserve(192 + name.size());
  uint32_t npairs = 16;
  mp_kv_uint(blob, "inode_id", 8, uint64_t(id));
  mp_str(blob, "name", 4);
  mp_str(blob, name);
  mp_kv_uint(blob, "file_type", 9, uint64_t(file_type));
  mp_kv_uint(blob, "length", 6, uint64_t(length));
  mp_str(blob, "is_complete", 11);
  blob.push_back(char(is_complete ? 0xc3 : 0xc2));
  mp_kv_uint(blob, "block_size", 10, uint64_t(block_size));
  mp_kv_uint(blob, "replicas", 8, uint64_t(replicas));
  mp_str(blob, "storage_tier", 12);
  mp_str(blob, storage_tier);
  mp_kv_uint(blob, "atime_ms", 8, uint64_t(atime_ms));
  mp_kv_uint(blob, "mode", 4, uint64_t(mode));
  mp_kv_uint(blob, "uid", 3, uint64_t(uid));
  mp_kv_uint(blob, "gid", 3, uint64_t(gid));
  mp_kv_uint(blob, "ttl_ms", 6, uint64_t(ttl_ms));
  mp_str(blob, "ttl_action", 10);
  mp_str(blob, ttl_action);
  mp_str(blob, "symlink_target", 14);
  mp_str(blob, symlink_target);
  mp_kv_uint(blob, "nlink", 5, uint64_t(nlink));
  if (!xattrs_blob.is_none()) {
    // pre-packed msgpack map under key "xattrs" (rare path)
    std::string xb = py::bytes(xattrs_blob);
    mp_str(blob, "xattrs", 6);
    blob += xb;
    npairs += 1;
  }
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  MetaNode& nd = S->nodes[id];
  nd.is_dir = is_dir;
  nd.npairs = npairs;
  nd.mtime = mtime_ms;
  nd.blob = std::move(blob);
  nd.blocks.clear();
  const int64_t* p = (const int64_t*)bb.data();
  size_t n = bb.size() / 16;
  nd.blocks.reserve(n);
  for (size_t i = 0; i < n; i++) nd.blocks.emplace_back(p[2 * i], p[2 * i + 1]);
}

// atomic node + children-edge insert for paged fault-ins: a concurrent
// native resolve must never observe a directory with a partially filled
// children map (that reads as FileNotFound).  children: repeated
// (u32 name_len LE, name bytes, i64 child_id LE).
static void meta_upsert_with_children(int64_t sid, int64_t id, bool is_dir,
                                      py::bytes blob, uint32_t npairs,
                                      py::bytes blocks, int64_t mtime,
                                      py::bytes children) {
  MetaServer* S = meta_get(sid);
  std::string b = blob;
  std::string bb = blocks;
  std::string cb = children;
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  MetaNode& nd = S->nodes[id];
  nd.is_dir = is_dir;
  nd.npairs = npairs;
  nd.mtime = mtime;
  nd.blob = std::move(b);
  nd.blocks.clear();
  const int64_t* p = (const int64_t*)bb.data();
  size_t n = bb.size() / 16;
  nd.blocks.reserve(n);
  for (size_t i = 0; i < n; i++) nd.blocks.emplace_back(p[2 * i], p[2 * i + 1]);
  nd.children.clear();
  const uint8_t* cp = (const uint8_t*)cb.data();
  const uint8_t* cend = cp + cb.size();
  while (cend - cp >= 4) {
    uint32_t ln = uint32_t(cp[0]) | (uint32_t(cp[1]) << 8) |
                  (uint32_t(cp[2]) << 16) | (uint32_t(cp[3]) << 24);
    cp += 4;
    if (size_t(cend - cp) < ln + 8) break;
    std::string name((const char*)cp, ln);
    cp += ln;
    int64_t cid;
    memcpy(&cid, cp, 8);
    cp += 8;
    nd.children[std::move(name)] = cid;
  }
}

static void meta_touch(int64_t sid, int64_t id, int64_t mtime) {
  MetaServer* S = meta_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  auto it = S->nodes.find(id);
  if (it != S->nodes.end()) it->second.mtime = mtime;
}

static void meta_add_child(int64_t sid, int64_t parent,
                           const std::string& name, int64_t child) {
  MetaServer* S = meta_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  auto it = S->nodes.find(parent);
  if (it != S->nodes.end()) it->second.children[name] = child;
}

static void meta_remove_child(int64_t sid, int64_t parent,
                              const std::string& name) {
  MetaServer* S = meta_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  auto it = S->nodes.find(parent);
  if (it != S->nodes.end()) it->second.children.erase(name);
}

static void meta_drop(int64_t sid, int64_t id) {
  MetaServer* S = meta_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  S->nodes.erase(id);
}

static void meta_clear(int64_t sid) {
  MetaServer* S = meta_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  S->nodes.clear();
}

static void meta_worker_upsert(int64_t sid, int64_t wid, py::bytes blob,
                               bool lost) {
  MetaServer* S = meta_get(sid);
  std::string b = blob;
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  WorkerRec& w = S->workers[wid];
  w.blob = std::move(b);
  w.lost = lost;
}

static void meta_block_add_loc(int64_t sid, int64_t bid, int64_t wid,
                               const std::string& tier, int ord) {
  MetaServer* S = meta_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  auto& v = S->block_locs[bid];
  for (auto& e : v) {
    if (e.wid == wid) {   // tier update (demotion) keeps position
      e.tier = tier;
      e.ord = uint8_t(ord);
      return;
    }
  }
  v.push_back(LocEnt{wid, uint8_t(ord), tier});
}

static void meta_block_remove_loc(int64_t sid, int64_t bid, int64_t wid) {
  MetaServer* S = meta_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  auto it = S->block_locs.find(bid);
  if (it == S->block_locs.end()) return;
  auto& v = it->second;
  v.erase(std::remove_if(v.begin(), v.end(),
                         [&](const LocEnt& e) { return e.wid == wid; }),
          v.end());
  if (v.empty()) S->block_locs.erase(it);
}

static void meta_block_drop(int64_t sid, int64_t bid) {
  MetaServer* S = meta_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->tree_mu);
  S->block_locs.erase(bid);
}

static py::dict meta_take_access(int64_t sid) {
  MetaServer* S = meta_get(sid);
  std::unordered_map<int64_t, uint64_t> taken;
  {
    std::lock_guard<std::mutex> g(S->acc_mu);
    taken.swap(S->access_counts);
  }
  py::dict d;
  for (auto& kv : taken) d[py::int_(kv.first)] = kv.second;
  return d;
}

static int meta_eventfd(int64_t sid) { return meta_get(sid)->fq_efd; }

static py::list meta_forward_pop(int64_t sid, int timeout_ms, int max_items) {
  MetaServer* S = meta_get(sid);
  std::vector<std::pair<uint64_t, std::string>> out;
  {
    py::gil_scoped_release rel;
    std::unique_lock<std::mutex> lk(S->fq_mu);
    if (S->fq.empty() && !S->stopping.load())
      S->fq_cv.wait_for(lk, std::chrono::milliseconds(timeout_ms));
    while (!S->fq.empty() && (int)out.size() < max_items) {
      out.emplace_back(std::move(S->fq.front()));
      S->fq.pop_front();
    }
  }
  py::list res;
  for (auto& it : out)
    res.append(py::make_tuple(it.first, py::bytes(it.second)));
  return res;
}

static bool meta_send(int64_t sid, uint64_t conn_id, py::bytes data) {
  MetaServer* S = meta_get(sid);
  std::shared_ptr<MetaConn> c;
  {
    std::lock_guard<std::mutex> g(S->conns_mu);
    auto it = S->conns.find(conn_id);
    if (it == S->conns.end()) return false;
    c = it->second;
  }
  std::string d = data;
  py::gil_scoped_release rel;
  return meta_write_all(c.get(), d.data(), d.size());
}

static py::dict meta_stats(int64_t sid) {
  MetaServer* S = meta_get(sid);
  py::dict d;
  d["served_status"] = S->served_status.load();
  d["served_list"] = S->served_list.load();
  d["served_exists"] = S->served_exists.load();
  d["served_open"] = S->served_open.load();
  d["served_ping"] = S->served_ping.load();
  d["served_notfound"] = S->served_notfound.load();
  d["forwarded"] = S->forwarded.load();
  d["conns_total"] = S->conns_total.load();
  {
    std::shared_lock<std::shared_mutex> lk(S->tree_mu);
    d["nodes"] = S->nodes.size();
    d["serving"] = S->serving;
  }
  return d;
}
