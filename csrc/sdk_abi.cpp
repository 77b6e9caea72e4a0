// C ABI for non-Python SDKs (the JNI binding target).
//
// The reference's Java SDK binds JNI symbols over `curvine-sdk-core`
// filesystem/reader/writer handles
// (/root/reference/crates/sdk/curvine-libsdk-java/src/java/java_abi.rs:25+,
// crates/sdk/curvine-sdk-core/src/lib_{filesystem,fs_reader,fs_writer}.rs).
// This exports the equivalent surface as plain `cv_*` C symbols from a
// pure C++ wire client (msgpack headers over the 22-byte frame protocol +
// the streaming worker data plane) — no interpreter anywhere — so a JNI
// (or Go/C#) shim is a thin drop-in.  Unit-tested from ctypes against a
// live MiniCluster (tests/test_c_abi.py).
//
// Handle model (mirrors lib_filesystem.rs / lib_fs_reader.rs /
// lib_fs_writer.rs): int64 handles for filesystem, reader, writer; every
// call returns 0 / positive on success, -errcode on failure;
// cv_last_error copies the thread-local message.
//
// Concurrency contract (same as the JNI SDK's usage): a handle is driven
// by one thread at a time; different handles may be used concurrently;
// close readers/writers before their filesystem handle.

#include <cstdarg>

// ------------------------------------------------------- msgpack DOM

struct CvVal {
  enum Kind { NIL, BOOL, INT, STR, BIN, ARR, MAP } kind = NIL;
  bool b = false;
  int64_t i = 0;
  std::string s;                       // STR/BIN payload
  std::vector<CvVal> arr;
  std::vector<std::pair<std::string, CvVal>> map;

  const CvVal* get(const char* key) const {
    for (auto& kv : map)
      if (kv.first == key) return &kv.second;
    return nullptr;
  }
  int64_t geti(const char* key, int64_t dflt = 0) const {
    const CvVal* v = get(key);
    return v ? v->i : dflt;
  }
  std::string gets(const char* key) const {
    const CvVal* v = get(key);
    return v ? v->s : std::string();
  }
};

static bool cv_parse(const uint8_t*& p, const uint8_t* end, CvVal* out,
                     int depth = 0) {
  if (p >= end || depth > 32) return false;
  uint8_t b = *p++;
  auto need = [&](size_t n) { return size_t(end - p) >= n; };
  if (b <= 0x7f) { out->kind = CvVal::INT; out->i = b; return true; }
  if (b >= 0xe0) { out->kind = CvVal::INT; out->i = int8_t(b); return true; }
  if (b >= 0xa0 && b <= 0xbf) {
    size_t n = b & 0x1f;
    if (!need(n)) return false;
    out->kind = CvVal::STR;
    out->s.assign((const char*)p, n);
    p += n;
    return true;
  }
  auto read_n = [&](size_t n) -> uint64_t {
    uint64_t v = 0;
    for (size_t i = 0; i < n; i++) v = (v << 8) | *p++;
    return v;
  };
  auto parse_arr = [&](size_t n) {
    out->kind = CvVal::ARR;
    out->arr.resize(n);
    for (size_t i = 0; i < n; i++)
      if (!cv_parse(p, end, &out->arr[i], depth + 1)) return false;
    return true;
  };
  auto parse_map = [&](size_t n) {
    out->kind = CvVal::MAP;
    out->map.resize(n);
    for (size_t i = 0; i < n; i++) {
      CvVal k;
      if (!cv_parse(p, end, &k, depth + 1)) return false;
      out->map[i].first = std::move(k.s);
      if (!cv_parse(p, end, &out->map[i].second, depth + 1)) return false;
    }
    return true;
  };
  if (b >= 0x80 && b <= 0x8f) return parse_map(b & 0xf);
  if (b >= 0x90 && b <= 0x9f) return parse_arr(b & 0xf);
  switch (b) {
    case 0xc0: out->kind = CvVal::NIL; return true;
    case 0xc2: out->kind = CvVal::BOOL; out->b = false; return true;
    case 0xc3: out->kind = CvVal::BOOL; out->b = true; return true;
    case 0xcc: if (!need(1)) return false;
      out->kind = CvVal::INT; out->i = int64_t(read_n(1)); return true;
    case 0xcd: if (!need(2)) return false;
      out->kind = CvVal::INT; out->i = int64_t(read_n(2)); return true;
    case 0xce: if (!need(4)) return false;
      out->kind = CvVal::INT; out->i = int64_t(read_n(4)); return true;
    case 0xcf: if (!need(8)) return false;
      out->kind = CvVal::INT; out->i = int64_t(read_n(8)); return true;
    case 0xd0: if (!need(1)) return false;
      out->kind = CvVal::INT; out->i = int8_t(read_n(1)); return true;
    case 0xd1: if (!need(2)) return false;
      out->kind = CvVal::INT; out->i = int16_t(read_n(2)); return true;
    case 0xd2: if (!need(4)) return false;
      out->kind = CvVal::INT; out->i = int32_t(read_n(4)); return true;
    case 0xd3: if (!need(8)) return false;
      out->kind = CvVal::INT; out->i = int64_t(read_n(8)); return true;
    case 0xd9: case 0xc4: {
      if (!need(1)) return false;
      size_t n = read_n(1);
      if (!need(n)) return false;
      out->kind = (b == 0xd9) ? CvVal::STR : CvVal::BIN;
      out->s.assign((const char*)p, n);
      p += n;
      return true;
    }
    case 0xda: case 0xc5: {
      if (!need(2)) return false;
      size_t n = read_n(2);
      if (!need(n)) return false;
      out->kind = (b == 0xda) ? CvVal::STR : CvVal::BIN;
      out->s.assign((const char*)p, n);
      p += n;
      return true;
    }
    case 0xdb: case 0xc6: {
      if (!need(4)) return false;
      size_t n = read_n(4);
      if (!need(n)) return false;
      out->kind = (b == 0xdb) ? CvVal::STR : CvVal::BIN;
      out->s.assign((const char*)p, n);
      p += n;
      return true;
    }
    case 0xdc: { if (!need(2)) return false; return parse_arr(read_n(2)); }
    case 0xdd: { if (!need(4)) return false; return parse_arr(read_n(4)); }
    case 0xde: { if (!need(2)) return false; return parse_map(read_n(2)); }
    case 0xdf: { if (!need(4)) return false; return parse_map(read_n(4)); }
    default: return false;
  }
}

// ------------------------------------------------------- header builder

struct CvMap {
  std::string buf;
  uint32_t n = 0;
  CvMap& kv_str(const char* k, const std::string& v) {
    mp_str(buf, k, strlen(k));
    mp_str(buf, v);
    n++;
    return *this;
  }
  CvMap& kv_int(const char* k, int64_t v) {
    mp_str(buf, k, strlen(k));
    if (v >= 0) {
      mp_uint(buf, uint64_t(v));
    } else {
      buf.push_back(char(0xd3));
      wr_u64be(buf, uint64_t(v));
    }
    n++;
    return *this;
  }
  CvMap& kv_bool(const char* k, bool v) {
    mp_str(buf, k, strlen(k));
    buf.push_back(char(v ? 0xc3 : 0xc2));
    n++;
    return *this;
  }
  std::string done() const {
    std::string o;
    if (n < 16) {
      o.push_back(char(0x80 | n));
    } else {
      o.push_back(char(0xde));
      wr_u16be(o, uint16_t(n));
    }
    o += buf;
    return o;
  }
};

// ------------------------------------------------------- error channel

static thread_local std::string g_cv_err;

static int cv_fail(int code, const char* fmt, ...) {
  char tmp[512];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(tmp, sizeof tmp, fmt, ap);
  va_end(ap);
  g_cv_err = tmp;
  return -code;
}

extern "C" int cv_last_error(char* buf, int cap) {
  if (cap <= 0) return 0;
  int n = int(std::min(g_cv_err.size(), size_t(cap - 1)));
  memcpy(buf, g_cv_err.data(), n);
  buf[n] = 0;
  return n;
}

// ------------------------------------------------------- filesystem

struct CvFilesystem {
  std::string host;
  int port = 0;
  int fd = -1;
  uint64_t next_req = 1;
  std::mutex mu;
  int64_t block_size = 64 << 20;
  std::string tier = "MEM";
};

struct CvReader {
  CvFilesystem* fs;
  int64_t length = 0;
  int64_t pos = 0;
  struct Blk {
    int64_t id, off, len;
    std::string host;
    int port;
  };
  std::vector<Blk> blocks;
};

struct CvWriter {
  CvFilesystem* fs;
  std::string path;
  int64_t block_size = 64 << 20;   // from the create reply
  int64_t pos = 0;
  // current block stream
  int dfd = -1;
  std::string dhost;
  int dport = 0;
  uint64_t req = 0;
  uint32_t seq = 0;
  int inflight = 0;
  int64_t cur_block = -1;
  int64_t cur_len = 0;
  int64_t cur_worker = -1;
  std::vector<int64_t> block_lens;
  std::vector<std::pair<int64_t, int64_t>> commits;  // (block_id, worker)
  bool failed = false;
};

static std::mutex g_cvh_mu;
static auto& g_cv_handles = *new std::unordered_map<int64_t, void*>();
static int64_t g_cvh_next = 1;

template <typename T>
static int64_t cvh_put(T* p) {
  std::lock_guard<std::mutex> g(g_cvh_mu);
  g_cv_handles[g_cvh_next] = p;
  return g_cvh_next++;
}

template <typename T>
static T* cvh_get(int64_t h) {
  std::lock_guard<std::mutex> g(g_cvh_mu);
  auto it = g_cv_handles.find(h);
  return it == g_cv_handles.end() ? nullptr : (T*)it->second;
}

static void cvh_del(int64_t h) {
  std::lock_guard<std::mutex> g(g_cvh_mu);
  g_cv_handles.erase(h);
}

// unary rpc; returns resp status nibble or -1, fills *out on success
static int cv_rpc(CvFilesystem* f, uint8_t code, const std::string& hdr,
                  CvVal* out) {
  std::lock_guard<std::mutex> g(f->mu);
  for (int attempt = 0; attempt < 2; attempt++) {
    if (f->fd < 0) f->fd = dc_connect(f->host, f->port);
    if (f->fd < 0) return -1;
    uint64_t req = f->next_req++;
    std::string frame = meta_proto(uint32_t(hdr.size()), 0, code, 0, req, 0);
    frame += hdr;
    if (!fd_write_all(f->fd, frame.data(), frame.size())) {
      close(f->fd);
      f->fd = -1;
      continue;
    }
    uint8_t proto[kMetaProto];
    if (!dc_read_exact(f->fd, proto, kMetaProto)) {
      close(f->fd);
      f->fd = -1;
      continue;
    }
    uint32_t hlen = rd_u32be(proto);
    uint32_t dlen = rd_u32be(proto + 4);
    std::vector<uint8_t> body(hlen + dlen);
    if ((hlen + dlen) && !dc_read_exact(f->fd, body.data(), hlen + dlen)) {
      close(f->fd);
      f->fd = -1;
      continue;
    }
    const uint8_t* p = body.data();
    if (hlen && !cv_parse(p, body.data() + hlen, out)) return -1;
    return proto[9] >> 4;
  }
  return -1;
}

static int cv_rpc_checked(CvFilesystem* f, uint8_t code,
                          const std::string& hdr, CvVal* out,
                          const char* what) {
  int st = cv_rpc(f, code, hdr, out);
  if (st < 0) return cv_fail(5, "%s: rpc transport failed", what);
  if (st == 5) {
    return cv_fail(int(out->geti("error_code", 1)), "%s: %s", what,
                   out->gets("error_msg").c_str());
  }
  return 0;
}

extern "C" int64_t cv_fs_new(const char* host, int port) {
  auto* f = new CvFilesystem();
  f->host = host;
  f->port = port;
  CvVal v;
  // probe liveness with a keepalive
  int st = cv_rpc(f, 1, std::string(), &v);
  if (st < 0) {
    delete f;
    cv_fail(5, "cv_fs_new: cannot reach master %s:%d", host, port);
    return -5;
  }
  return cvh_put(f);
}

extern "C" int cv_fs_close(int64_t h) {
  CvFilesystem* f = cvh_get<CvFilesystem>(h);
  if (!f) return -22;
  cvh_del(h);
  if (f->fd >= 0) close(f->fd);
  delete f;
  return 0;
}

extern "C" int cv_mkdir(int64_t h, const char* path) {
  CvFilesystem* f = cvh_get<CvFilesystem>(h);
  if (!f) return -22;
  CvVal v;
  return cv_rpc_checked(
      f, 2, CvMap().kv_str("path", path).kv_bool("create_parents", true)
                .done(), &v, "mkdir");
}

struct CvStatus {  // mirrors FileStatusProto essentials (C layout)
  int64_t inode_id;
  int64_t length;
  int64_t mtime_ms;
  int32_t file_type;   // 0 file, 1 dir, 2 symlink
  int32_t is_complete;
  int32_t mode;
  int32_t nlink;
};

static void fill_status(const CvVal& st, CvStatus* out) {
  out->inode_id = st.geti("inode_id");
  out->length = st.geti("length");
  out->mtime_ms = st.geti("mtime_ms");
  out->file_type = int32_t(st.geti("file_type"));
  const CvVal* c = st.get("is_complete");
  out->is_complete = c && (c->kind == CvVal::BOOL ? c->b : c->i != 0);
  out->mode = int32_t(st.geti("mode", 0644));
  out->nlink = int32_t(st.geti("nlink", 1));
}

extern "C" int cv_get_status(int64_t h, const char* path, CvStatus* out) {
  CvFilesystem* f = cvh_get<CvFilesystem>(h);
  if (!f) return -22;
  CvVal v;
  int rc = cv_rpc_checked(f, 7, CvMap().kv_str("path", path).done(), &v,
                          "get_status");
  if (rc) return rc;
  const CvVal* st = v.get("status");
  if (!st) return cv_fail(5, "get_status: malformed reply");
  fill_status(*st, out);
  return 0;
}

extern "C" int cv_exists(int64_t h, const char* path) {
  CvFilesystem* f = cvh_get<CvFilesystem>(h);
  if (!f) return -22;
  CvVal v;
  int rc = cv_rpc_checked(f, 9, CvMap().kv_str("path", path).done(), &v,
                          "exists");
  if (rc) return rc;
  const CvVal* e = v.get("exists");
  return (e && e->b) ? 1 : 0;
}

extern "C" int cv_rename(int64_t h, const char* src, const char* dst) {
  CvFilesystem* f = cvh_get<CvFilesystem>(h);
  if (!f) return -22;
  CvVal v;
  return cv_rpc_checked(
      f, 10, CvMap().kv_str("src", src).kv_str("dst", dst).done(), &v,
      "rename");
}

extern "C" int cv_delete(int64_t h, const char* path, int recursive) {
  CvFilesystem* f = cvh_get<CvFilesystem>(h);
  if (!f) return -22;
  CvVal v;
  return cv_rpc_checked(
      f, 3,
      CvMap().kv_str("path", path).kv_bool("recursive", recursive != 0)
          .done(), &v, "delete");
}

// newline-joined child names into buf; returns bytes written or -err
extern "C" int cv_list_status(int64_t h, const char* path, char* buf,
                              int cap) {
  CvFilesystem* f = cvh_get<CvFilesystem>(h);
  if (!f) return -22;
  CvVal v;
  int rc = cv_rpc_checked(f, 8, CvMap().kv_str("path", path).done(), &v,
                          "list_status");
  if (rc) return rc;
  const CvVal* sts = v.get("statuses");
  if (!sts) return cv_fail(5, "list_status: malformed reply");
  std::string out;
  for (auto& st : sts->arr) {
    if (!out.empty()) out.push_back('\n');
    out += st.gets("name");
  }
  int n = int(std::min(out.size(), size_t(cap - 1)));
  memcpy(buf, out.data(), n);
  buf[n] = 0;
  return n;
}

// ------------------------------------------------------- reader

extern "C" int64_t cv_open(int64_t h, const char* path) {
  CvFilesystem* f = cvh_get<CvFilesystem>(h);
  if (!f) return -22;
  CvVal v;
  int rc = cv_rpc_checked(f, 5, CvMap().kv_str("path", path).done(), &v,
                          "open");
  if (rc) return rc;
  const CvVal* fb = v.get("file_blocks");
  if (!fb) return cv_fail(5, "open: malformed reply");
  auto* r = new CvReader();
  r->fs = f;
  const CvVal* st = fb->get("status");
  r->length = st ? st->geti("length") : 0;
  const CvVal* blocks = fb->get("blocks");
  if (blocks) {
    for (auto& lb : blocks->arr) {
      const CvVal* b = lb.get("block");
      const CvVal* locs = lb.get("locations");
      if (!b || !locs || locs->arr.empty()) continue;
      CvReader::Blk blk;
      blk.id = b->geti("block_id");
      blk.len = b->geti("length");
      blk.off = lb.geti("offset");
      blk.host = locs->arr[0].gets("hostname");
      blk.port = int(locs->arr[0].geti("rpc_port"));
      r->blocks.push_back(std::move(blk));
    }
  }
  return cvh_put(r);
}

// one-shot data-plane block read (Open -> frames -> Complete)
static int64_t cv_read_block(const std::string& host, int port, int64_t bid,
                             int64_t off, int64_t len, uint8_t* out) {
  int fd = dc_acquire(host, port);
  if (fd < 0) return -1;
  bool keep = false;
  int64_t got = 0;
  uint64_t req = g_dc_req.fetch_add(1);
  std::string hmap = CvMap().kv_int("block_id", bid).kv_int("offset", off)
                         .kv_int("length", len)
                         .kv_int("chunk_size", 4 << 20).done();
  std::string frame =
      meta_proto(uint32_t(hmap.size()), 0, kCodeReadBlock, 1, req, 0);
  frame += hmap;
  if (fd_write_all(fd, frame.data(), frame.size())) {
    for (;;) {
      uint8_t proto[kMetaProto];
      if (!dc_read_exact(fd, proto, kMetaProto)) break;
      uint32_t hlen = rd_u32be(proto);
      uint32_t dlen = rd_u32be(proto + 4);
      if (hlen > kMetaMaxLen || dlen > kMetaMaxLen) break;
      std::vector<uint8_t> hdr(hlen);
      if (hlen && !dc_read_exact(fd, hdr.data(), hlen)) break;
      if (dlen) {
        if (got + dlen > len) break;
        if (!dc_read_exact(fd, out + got, dlen)) break;
        got += dlen;
      }
      uint8_t st = proto[9] >> 4;
      if (st == 3) { keep = true; break; }
      if (st == 5) break;
    }
  }
  dc_release(host, port, fd, keep);
  return keep ? got : -1;
}

extern "C" int64_t cv_read(int64_t rh, void* buf, int64_t n) {
  CvReader* r = cvh_get<CvReader>(rh);
  if (!r) return -22;
  n = std::min(n, r->length - r->pos);
  if (n <= 0) return 0;
  uint8_t* out = (uint8_t*)buf;
  int64_t got = 0;
  while (got < n) {
    // locate the block containing pos
    const CvReader::Blk* blk = nullptr;
    for (auto& b : r->blocks)
      if (r->pos >= b.off && r->pos < b.off + b.len) { blk = &b; break; }
    if (!blk) break;
    int64_t boff = r->pos - blk->off;
    int64_t want = std::min(n - got, blk->len - boff);
    int64_t k = cv_read_block(blk->host, blk->port, blk->id, boff, want,
                              out + got);
    if (k <= 0) {
      if (got) break;
      return cv_fail(5, "cv_read: block %lld stream failed",
                     (long long)blk->id);
    }
    got += k;
    r->pos += k;
  }
  return got;
}

extern "C" int cv_seek(int64_t rh, int64_t pos) {
  CvReader* r = cvh_get<CvReader>(rh);
  if (!r) return -22;
  if (pos < 0 || pos > r->length) return -22;
  r->pos = pos;
  return 0;
}

extern "C" int64_t cv_reader_len(int64_t rh) {
  CvReader* r = cvh_get<CvReader>(rh);
  return r ? r->length : -22;
}

extern "C" int cv_close_reader(int64_t rh) {
  CvReader* r = cvh_get<CvReader>(rh);
  if (!r) return -22;
  cvh_del(rh);
  delete r;
  return 0;
}

// ------------------------------------------------------- writer

extern "C" int64_t cv_create(int64_t h, const char* path, int overwrite) {
  CvFilesystem* f = cvh_get<CvFilesystem>(h);
  if (!f) return -22;
  CvVal v;
  int rc = cv_rpc_checked(
      f, 4,
      CvMap().kv_str("path", path).kv_bool("overwrite", overwrite != 0)
          .done(), &v, "create");
  if (rc) return rc;
  auto* w = new CvWriter();
  w->fs = f;
  w->path = path;
  const CvVal* st = v.get("status");
  if (st) {
    int64_t bs = st->geti("block_size");
    if (bs > 0) w->block_size = bs;
  }
  return cvh_put(w);
}

static bool cvw_ack(CvWriter* w) {
  uint8_t proto[kMetaProto];
  if (!dc_read_exact(w->dfd, proto, kMetaProto)) return false;
  uint32_t hlen = rd_u32be(proto);
  uint32_t dlen = rd_u32be(proto + 4);
  std::vector<uint8_t> tmp(hlen + dlen);
  if ((hlen + dlen) && !dc_read_exact(w->dfd, tmp.data(), hlen + dlen))
    return false;
  return (proto[9] >> 4) != 5;
}

static int cvw_finish_block(CvWriter* w) {
  if (w->dfd < 0) return 0;
  bool ok = true;
  while (ok && w->inflight > 0) {
    ok = cvw_ack(w);
    w->inflight--;
  }
  if (ok) {
    std::string fh = CvMap().kv_int("length", w->cur_len).done();
    std::string cf = meta_proto(uint32_t(fh.size()), 0, kCodeWriteBlock, 3,
                                w->req, w->seq++);
    cf += fh;
    ok = fd_write_all(w->dfd, cf.data(), cf.size()) && cvw_ack(w);
  }
  close(w->dfd);
  w->dfd = -1;
  if (!ok) {
    w->failed = true;
    return cv_fail(5, "write: block %lld commit failed",
                   (long long)w->cur_block);
  }
  w->block_lens.push_back(w->cur_len);
  w->commits.emplace_back(w->cur_block, w->cur_worker);
  w->cur_block = -1;
  w->cur_len = 0;
  return 0;
}

static int cvw_open_block(CvWriter* w) {
  CvVal v;
  int rc = cv_rpc_checked(w->fs, 11,
                          CvMap().kv_str("path", w->path).done(), &v,
                          "add_block");
  if (rc) return rc;
  const CvVal* lb = v.get("block");
  const CvVal* b = lb ? lb->get("block") : nullptr;
  const CvVal* locs = lb ? lb->get("locations") : nullptr;
  if (!b || !locs || locs->arr.empty())
    return cv_fail(5, "add_block: malformed reply");
  w->cur_block = b->geti("block_id");
  w->cur_worker = locs->arr[0].geti("worker_id");
  w->dhost = locs->arr[0].gets("hostname");
  w->dport = int(locs->arr[0].geti("rpc_port"));
  w->dfd = dc_connect(w->dhost, w->dport);
  if (w->dfd < 0) return cv_fail(5, "worker connect failed");
  w->req = g_dc_req.fetch_add(1);
  w->seq = 0;
  w->inflight = 0;
  std::string hmap = CvMap().kv_int("block_id", w->cur_block)
                         .kv_int("reserve", w->block_size)
                         .kv_str("tier", w->fs->tier).done();
  std::string frame =
      meta_proto(uint32_t(hmap.size()), 0, kCodeWriteBlock, 1, w->req,
                 w->seq++);
  frame += hmap;
  if (!fd_write_all(w->dfd, frame.data(), frame.size()) || !cvw_ack(w)) {
    close(w->dfd);
    w->dfd = -1;
    return cv_fail(5, "write open failed");
  }
  return 0;
}

extern "C" int64_t cv_write(int64_t wh, const void* buf, int64_t n) {
  CvWriter* w = cvh_get<CvWriter>(wh);
  if (!w) return -22;
  if (w->failed) return -5;
  const uint8_t* p = (const uint8_t*)buf;
  int64_t left = n;
  while (left > 0) {
    if (w->dfd < 0) {
      int rc = cvw_open_block(w);
      if (rc) return rc;
    }
    int64_t room = w->block_size - w->cur_len;
    int64_t take = std::min<int64_t>({left, room, 4 << 20});
    std::string ph = meta_proto(0, uint32_t(take), kCodeWriteBlock, 2,
                                w->req, w->seq++);
    struct iovec iov[2] = {{(void*)ph.data(), ph.size()},
                           {(void*)p, size_t(take)}};
    struct msghdr mh = {};
    mh.msg_iov = iov;
    mh.msg_iovlen = 2;
    size_t sent = 0, total = ph.size() + size_t(take);
    while (sent < total) {
      ssize_t k = sendmsg(w->dfd, &mh, MSG_NOSIGNAL);
      if (k < 0) {
        if (errno == EINTR) continue;
        w->failed = true;
        return cv_fail(5, "cv_write: send failed");
      }
      sent += size_t(k);
      size_t adv = size_t(k);
      for (int i = 0; i < 2 && adv; i++) {
        size_t t = std::min(adv, iov[i].iov_len);
        iov[i].iov_base = (char*)iov[i].iov_base + t;
        iov[i].iov_len -= t;
        adv -= t;
      }
      while (mh.msg_iovlen && mh.msg_iov->iov_len == 0) {
        mh.msg_iov++;
        mh.msg_iovlen--;
      }
    }
    w->inflight++;
    while (w->inflight >= 4) {
      if (!cvw_ack(w)) {
        w->failed = true;
        return cv_fail(5, "cv_write: worker rejected chunk");
      }
      w->inflight--;
    }
    p += take;
    left -= take;
    w->cur_len += take;
    w->pos += take;
    if (w->cur_len >= w->block_size) {
      int rc = cvw_finish_block(w);
      if (rc) return rc;
    }
  }
  return n;
}

extern "C" int cv_close_writer(int64_t wh) {
  CvWriter* w = cvh_get<CvWriter>(wh);
  if (!w) return -22;
  int rc = 0;
  if (!w->failed) {
    rc = cvw_finish_block(w);
    if (!rc) {
      CvMap m;
      m.kv_str("path", w->path).kv_int("length", w->pos);
      // block_lens array
      mp_str(m.buf, "block_lens", 10);
      size_t nb = w->block_lens.size();
      if (nb < 16) {
        m.buf.push_back(char(0x90 | nb));
      } else {
        m.buf.push_back(char(0xdc));
        wr_u16be(m.buf, uint16_t(nb));
      }
      for (int64_t l : w->block_lens) mp_uint(m.buf, uint64_t(l));
      m.n++;
      // commits: [{block_id, locations:[wid], tiers:[tier]}...]
      mp_str(m.buf, "commits", 7);
      if (nb < 16) {
        m.buf.push_back(char(0x90 | nb));
      } else {
        m.buf.push_back(char(0xdc));
        wr_u16be(m.buf, uint16_t(nb));
      }
      for (auto& c : w->commits) {
        m.buf.push_back(char(0x83));
        mp_str(m.buf, "block_id", 8);
        mp_uint(m.buf, uint64_t(c.first));
        mp_str(m.buf, "locations", 9);
        m.buf.push_back(char(0x91));
        mp_uint(m.buf, uint64_t(c.second));
        mp_str(m.buf, "tiers", 5);
        m.buf.push_back(char(0x91));
        mp_str(m.buf, w->fs->tier);
      }
      m.n++;
      CvVal v;
      rc = cv_rpc_checked(w->fs, 12, m.done(), &v, "complete_file");
    }
  } else {
    rc = -5;
  }
  cvh_del(wh);
  if (w->dfd >= 0) close(w->dfd);
  delete w;
  return rc;
}
