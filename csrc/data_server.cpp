// Native worker streaming data plane: GIL-free epoll threads own the
// worker's client sockets and serve block READ streams straight from the
// arenas/files (zero Python in the hot loop), consume block WRITE data
// frames directly into the reserved extents, and forward every control
// frame (stream Open/Complete, heartbeats, short-circuit info) to the
// Python WorkerHandler.
//
// This is the MI355X answer to the reference's splice/sendfile worker hot
// path (crates/core/rpc/src/handler/rpc_frame.rs:82-148 send-side
// sendfile loop, curvine-worker read_handler.rs:183-214 chunk streaming):
//  * MEM-tier (host/pinned arena) blocks: sendmsg() straight from the
//    arena base — no staging copy at all;
//  * HBM-tier blocks: double-buffered hipMemcpyAsync D2H into pinned
//    bounce buffers overlapping the socket sends;
//  * SSD/HDD file blocks: sendfile(2) from the block file to the socket;
//  * write data frames: received into the connection buffer once, then
//    memcpy/H2D straight into the reserved arena extent or pwrite(2) to
//    the block file — the Python handler only sees Open/Complete.
//
// Shares the wire helpers (mp_*, rd_*, wr_*) and the Arena machinery with
// the rest of the translation unit (this file is #include'd from
// module.cpp after meta_server.cpp).

#include <netdb.h>
#include <sys/sendfile.h>
#include <sys/types.h>
#include <sys/uio.h>

// ---------------------------------------------------------------- header scan

// scan a top-level msgpack map for an integer field; returns false if absent
static bool mp_find_int(const uint8_t* p, const uint8_t* end, const char* key,
                        size_t keylen, int64_t* out) {
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
    if (kn == keylen && memcmp(ks, key, keylen) == 0) {
      if (p >= end) return false;
      uint8_t v = *p;
      if (v <= 0x7f) { *out = v; p++; return true; }
      if (v >= 0xe0) { *out = int8_t(v); p++; return true; }
      p++;
      switch (v) {
        case 0xcc: if (p >= end) return false; *out = *p; return true;
        case 0xcd: if (end - p < 2) return false;
          *out = (int64_t(p[0]) << 8) | p[1]; return true;
        case 0xce: if (end - p < 4) return false;
          *out = rd_u32be(p); return true;
        case 0xcf: if (end - p < 8) return false;
          *out = int64_t(rd_u64be(p)); return true;
        case 0xd0: if (p >= end) return false; *out = int8_t(*p); return true;
        case 0xd1: if (end - p < 2) return false;
          *out = int16_t((p[0] << 8) | p[1]); return true;
        case 0xd2: if (end - p < 4) return false;
          *out = int32_t(rd_u32be(p)); return true;
        case 0xd3: if (end - p < 8) return false;
          *out = int64_t(rd_u64be(p)); return true;
        default: return false;
      }
    }
    if (!mp_skip(p, end)) return false;
  }
  return false;
}

// ---------------------------------------------------------------- state

static constexpr uint8_t kCodeWriteBlock = 80;
static constexpr uint8_t kCodeReadBlock = 81;

struct DataBlock {
  int kind = 0;  // 0 = arena extent, 1 = file
  int arena = -1;
  uint64_t aoff = 0;
  int64_t len = 0;
  std::string path;
  bool direct = false;   // file kind: serve with O_DIRECT aligned preads
  std::atomic<int> refs{0};
  std::atomic<bool> dead{false};
};

struct WriteSess {
  int kind = 0;  // 0 = arena, 1 = file
  int arena = -1;
  uint64_t aoff = 0;
  uint64_t reserved = 0;
  int fd = -1;                  // file kind: dup of the layout's open fd
  std::atomic<uint64_t> pos{0};  // append watermark (finalize default len)
};

// Receive buffer that can be upgraded to PINNED host memory: once an
// HBM write session opens on a conn, frames land in hipHostMalloc'd
// pages so dev_write DMAs straight from the receive buffer (no staging
// memcpy — the socket recv is the only host copy).
struct RBuf {
  uint8_t* p = nullptr;
  size_t cap = 0;
  bool pinned = false;
  std::atomic<bool> want_pinned{false};   // set from the register path

  uint8_t* data() { return p; }
  size_t capacity() const { return cap; }

  void grow(size_t need, size_t keep) {
    bool wp = want_pinned.load(std::memory_order_relaxed);
    bool to_pin = wp && !pinned;
    if (need <= cap && !to_pin) return;
    size_t ncap = cap ? cap : 256u << 10;
    while (ncap < need) ncap *= 2;
    uint8_t* np = nullptr;
    bool np_pinned = false;
    if (wp &&
        hipHostMalloc((void**)&np, ncap, hipHostMallocDefault) ==
            hipSuccess) {
      np_pinned = true;
    } else {
      (void)hipGetLastError();
      np = (uint8_t*)std::malloc(ncap);
      if (!np) throw std::bad_alloc();
    }
    if (keep && p) std::memcpy(np, p, keep);
    release();
    p = np;
    cap = ncap;
    pinned = np_pinned;
  }

  void release() {
    if (!p) return;
    if (pinned) (void)hipHostFree(p);
    else std::free(p);
    p = nullptr;
    cap = 0;
    pinned = false;
  }

  ~RBuf() { release(); }
};

struct DataConn {
  int fd = -1;
  uint64_t id = 0;
  RBuf rbuf;
  size_t rlen = 0;
  std::mutex wmu;
  std::atomic<bool> dead{false};
  std::mutex smu;
  std::unordered_map<uint64_t, std::shared_ptr<WriteSess>> sess;
};

struct DataServer {
  int listen_fd = -1, epfd = -1;
  std::atomic<bool> stopping{false};
  std::shared_mutex blk_mu;
  std::unordered_map<int64_t, std::shared_ptr<DataBlock>> blocks;
  // dropped blocks with in-flight readers, polled via data_block_refs
  std::unordered_map<int64_t, std::shared_ptr<DataBlock>> dying;
  std::mutex conns_mu;
  std::unordered_map<uint64_t, std::shared_ptr<DataConn>> conns;
  std::atomic<uint64_t> next_conn{1};
  std::vector<std::thread> threads;
  std::mutex fq_mu;
  std::condition_variable fq_cv;
  std::deque<std::pair<uint64_t, std::string>> fq;
  int fq_efd = -1;
  std::atomic<uint64_t> served_reads{0}, served_read_bytes{0},
      served_writes{0}, served_write_bytes{0}, forwarded{0}, conns_total{0};
};

static std::mutex g_data_mu;
static auto& g_data =
    *new std::unordered_map<int64_t, std::unique_ptr<DataServer>>();
static int64_t g_data_next = 1;

static DataServer* data_get(int64_t sid) {
  std::lock_guard<std::mutex> g(g_data_mu);
  auto it = g_data.find(sid);
  if (it == g_data.end()) throw std::runtime_error("bad data server id");
  return it->second.get();
}

// ---------------------------------------------------------------- io helpers

static bool fd_write_all(int fd, const char* p, size_t n) {
  while (n) {
    ssize_t w = send(fd, p, n, MSG_NOSIGNAL);
    if (w > 0) { p += w; n -= size_t(w); continue; }
    if (w < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
      struct pollfd pf = {fd, POLLOUT, 0};
      if (poll(&pf, 1, 30000) <= 0) return false;
      continue;
    }
    if (w < 0 && errno == EINTR) continue;
    return false;
  }
  return true;
}

// one frame = proto(+header) then payload from a flat pointer; wmu held by
// caller so stream frames never interleave with forwarded Python replies
static bool frame_send_locked(int fd, const std::string& head,
                              const uint8_t* payload, size_t n) {
  if (n) {
    struct iovec iov[2] = {{(void*)head.data(), head.size()},
                           {(void*)payload, n}};
    struct msghdr mh = {};
    mh.msg_iov = iov;
    mh.msg_iovlen = 2;
    size_t sent = 0, total = head.size() + n;
    while (sent < total) {
      ssize_t w = sendmsg(fd, &mh, MSG_NOSIGNAL);
      if (w < 0) {
        if (errno == EINTR) continue;
        if (errno == EAGAIN || errno == EWOULDBLOCK) {
          struct pollfd pf = {fd, POLLOUT, 0};
          if (poll(&pf, 1, 30000) <= 0) return false;
          continue;
        }
        return false;
      }
      sent += size_t(w);
      // advance iov
      size_t adv = size_t(w);
      for (int i = 0; i < 2 && adv; i++) {
        size_t take = std::min(adv, iov[i].iov_len);
        iov[i].iov_base = (char*)iov[i].iov_base + take;
        iov[i].iov_len -= take;
        adv -= take;
      }
      while (mh.msg_iovlen && mh.msg_iov->iov_len == 0) {
        mh.msg_iov++;
        mh.msg_iovlen--;
      }
    }
    return true;
  }
  return fd_write_all(fd, head.data(), head.size());
}

static bool frame_sendfile_locked(int sock, const std::string& head, int fd,
                                  off_t off, size_t n) {
  if (!fd_write_all(sock, head.data(), head.size())) return false;
  while (n) {
    ssize_t w = sendfile(sock, fd, &off, n);
    if (w > 0) { n -= size_t(w); continue; }
    if (w < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
      struct pollfd pf = {sock, POLLOUT, 0};
      if (poll(&pf, 1, 30000) <= 0) return false;
      continue;
    }
    if (w < 0 && errno == EINTR) continue;
    return false;
  }
  return true;
}

static void data_close_conn(DataServer* S, const std::shared_ptr<DataConn>& c) {
  bool was = c->dead.exchange(true);
  if (was) return;
  epoll_ctl(S->epfd, EPOLL_CTL_DEL, c->fd, nullptr);
  close(c->fd);
  {
    std::lock_guard<std::mutex> g(c->smu);
    for (auto& kv : c->sess)
      if (kv.second->fd >= 0) close(kv.second->fd);
    c->sess.clear();
  }
  {
    std::lock_guard<std::mutex> g(S->conns_mu);
    S->conns.erase(c->id);
  }
  {
    std::lock_guard<std::mutex> g(S->fq_mu);
    S->fq.emplace_back(c->id, std::string());  // close sentinel for Python
  }
  S->fq_cv.notify_one();
  if (S->fq_efd >= 0) {
    uint64_t one = 1;
    ssize_t r = write(S->fq_efd, &one, 8);
    (void)r;
  }
}

static void data_forward(DataServer* S, const std::shared_ptr<DataConn>& c,
                         const uint8_t* frame, size_t total) {
  S->forwarded.fetch_add(1, std::memory_order_relaxed);
  bool was_empty;
  {
    std::lock_guard<std::mutex> g(S->fq_mu);
    was_empty = S->fq.empty();
    S->fq.emplace_back(c->id, std::string((const char*)frame, total));
  }
  S->fq_cv.notify_one();
  if (was_empty && S->fq_efd >= 0) {
    uint64_t one = 1;
    ssize_t r = write(S->fq_efd, &one, 8);
    (void)r;
  }
}

// The pinned bounce pool (BouncePair/bounce_acquire/bounce_release) is
// shared module infrastructure — defined in module.cpp ahead of the
// arena copy engines, used here for HBM stream serving.

// ---------------------------------------------------------------- read serve

// false -> forward the frame to Python (unknown block / torn state)
static bool data_serve_read(DataServer* S, DataConn* c, const uint8_t* frame,
                            uint32_t hlen, uint8_t status, uint64_t req_id,
                            uint32_t seq) {
  const uint8_t* h = frame + kMetaProto;
  const uint8_t* hend = h + hlen;
  int64_t block_id = -1, offset = 0, length = -1, chunk = 1 << 20;
  if (!mp_find_int(h, hend, "block_id", 8, &block_id)) return false;
  mp_find_int(h, hend, "offset", 6, &offset);
  mp_find_int(h, hend, "length", 6, &length);
  mp_find_int(h, hend, "chunk_size", 10, &chunk);
  if (chunk <= 0 || chunk > int64_t(kMetaMaxLen)) chunk = 1 << 20;

  std::shared_ptr<DataBlock> b;
  {
    std::shared_lock<std::shared_mutex> lk(S->blk_mu);
    auto it = S->blocks.find(block_id);
    if (it == S->blocks.end()) return false;
    b = it->second;
  }
  if (b->dead.load()) return false;
  b->refs.fetch_add(1);
  struct RefGuard {
    DataBlock* b;
    ~RefGuard() { b->refs.fetch_sub(1); }
  } rg{b.get()};

  uint8_t req_status = status & 0xF;
  if (offset < 0) offset = 0;
  int64_t avail = b->len - offset;
  if (avail < 0) avail = 0;
  int64_t n = (length < 0) ? avail : std::min<int64_t>(length, avail);

  // open-ack: Running reply with the block length
  std::string ack_hdr;
  ack_hdr.push_back(char(0x81));
  mp_str(ack_hdr, "length", 6);
  mp_uint(ack_hdr, uint64_t(b->len));
  std::string head = meta_proto(uint32_t(ack_hdr.size()), 0, kCodeReadBlock,
                                uint8_t((2 << 4) | req_status), req_id, seq);
  head += ack_hdr;

  std::lock_guard<std::mutex> wg(c->wmu);
  if (c->dead.load()) return true;
  if (!fd_write_all(c->fd, head.data(), head.size())) {
    c->dead.store(true);
    return true;
  }

  bool ok = true;
  if (b->kind == 0) {
    Arena* a = get_arena(b->arena);
    if (!a->is_dev()) {
      const uint8_t* src = (const uint8_t*)a->base + b->aoff + offset;
      int64_t pos = 0;
      while (pos < n && ok) {
        int64_t cn = std::min<int64_t>(chunk, n - pos);
        std::string ph = meta_proto(0, uint32_t(cn), kCodeReadBlock,
                                    uint8_t((2 << 4) | req_status), req_id,
                                    seq);
        ok = frame_send_locked(c->fd, ph, src + pos, size_t(cn));
        pos += cn;
      }
    } else {
      // HBM: double-buffered D2H into pooled pinned bounce buffers,
      // overlapping the socket sends
      HIP_CHECK(hipSetDevice(a->device));
      hipStream_t s = thread_stream(a->device);
      size_t bsz = std::min<int64_t>(chunk, kBounceSz);
      BouncePair* bp = bounce_acquire(a->device);
      struct BounceGuard {
        BouncePair* b;
        hipStream_t s;
        ~BounceGuard() {
          // an exception can leave an async D2H in flight targeting the
          // pinned buffers: drain before the pair goes back to the pool
          hipStreamSynchronize(s);
          bounce_release(b);
        }
      } bg{bp, s};
      void** pin = bp->pin;
      hipEvent_t* ev = bp->ev;
      const uint8_t* src = (const uint8_t*)a->base + b->aoff + offset;
      int64_t nchunks = (n + bsz - 1) / int64_t(bsz);
      for (int64_t k = 0; k < nchunks && ok; k++) {
        int64_t coff = k * int64_t(bsz);
        int64_t cn = std::min<int64_t>(bsz, n - coff);
        int slot = int(k & 1);
        HIP_CHECK(hipMemcpyAsync(pin[slot], src + coff, cn,
                                 hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipEventRecord(ev[slot], s));
        if (k > 0) {
          // previous chunk's copy has completed by FIFO stream order;
          // but sync its event explicitly before touching the buffer
          int prev = int((k - 1) & 1);
          HIP_CHECK(hipEventSynchronize(ev[prev]));
          int64_t poff = (k - 1) * int64_t(bsz);
          int64_t pn = std::min<int64_t>(bsz, n - poff);
          std::string ph = meta_proto(0, uint32_t(pn), kCodeReadBlock,
                                      uint8_t((2 << 4) | req_status), req_id,
                                      seq);
          ok = frame_send_locked(c->fd, ph, (const uint8_t*)pin[prev],
                                 size_t(pn));
        }
      }
      if (ok && nchunks > 0) {
        int last = int((nchunks - 1) & 1);
        HIP_CHECK(hipEventSynchronize(ev[last]));
        int64_t poff = (nchunks - 1) * int64_t(bsz);
        int64_t pn = n - poff;
        std::string ph = meta_proto(0, uint32_t(pn), kCodeReadBlock,
                                    uint8_t((2 << 4) | req_status), req_id,
                                    seq);
        ok = frame_send_locked(c->fd, ph, (const uint8_t*)pin[last],
                               size_t(pn));
      }
    }
  } else if (b->direct) {
    // NVMe page-cache bypass: aligned O_DIRECT preads into an aligned
    // bounce, sent with sendmsg (sendfile needs the page cache)
    int fd = open(b->path.c_str(), O_RDONLY | O_DIRECT);
    if (fd < 0) fd = open(b->path.c_str(), O_RDONLY);  // fs w/o DIO
    if (fd < 0) {
      ok = false;
    } else {
      constexpr int64_t kAlign = 4096;
      size_t bsz = size_t(std::min<int64_t>(chunk + 2 * kAlign, 16 << 20));
      void* abuf = nullptr;
      if (posix_memalign(&abuf, kAlign, bsz) != 0) abuf = nullptr;
      int64_t pos = 0;
      while (abuf && pos < n && ok) {
        int64_t want = std::min<int64_t>(chunk, n - pos);
        int64_t fo = offset + pos;
        int64_t lo = fo & ~(kAlign - 1);
        int64_t span = fo + want - lo;
        span = (span + kAlign - 1) & ~(kAlign - 1);
        int64_t got = 0;
        while (got < span) {
          ssize_t r = pread(fd, (uint8_t*)abuf + got, size_t(span - got),
                            off_t(lo + got));
          if (r < 0 && errno == EINTR) continue;
          if (r <= 0) break;   // EOF tail (unaligned file end)
          got += r;
        }
        int64_t have = std::min<int64_t>(want, got - (fo - lo));
        if (have < want) { ok = false; break; }
        std::string ph = meta_proto(0, uint32_t(want), kCodeReadBlock,
                                    uint8_t((2 << 4) | req_status), req_id,
                                    seq);
        ok = frame_send_locked(c->fd, ph,
                               (const uint8_t*)abuf + (fo - lo),
                               size_t(want));
        pos += want;
      }
      if (!abuf) ok = false;
      free(abuf);
      close(fd);
    }
  } else {
    int fd = open(b->path.c_str(), O_RDONLY);
    if (fd < 0) {
      ok = false;
    } else {
      int64_t pos = 0;
      while (pos < n && ok) {
        int64_t cn = std::min<int64_t>(chunk, n - pos);
        std::string ph = meta_proto(0, uint32_t(cn), kCodeReadBlock,
                                    uint8_t((2 << 4) | req_status), req_id,
                                    seq);
        ok = frame_sendfile_locked(c->fd, ph, fd, off_t(offset + pos),
                                   size_t(cn));
        pos += cn;
      }
      close(fd);
    }
  }
  if (ok) {
    std::string done = meta_proto(0, 0, kCodeReadBlock,
                                  uint8_t((3 << 4) | req_status), req_id, seq);
    ok = fd_write_all(c->fd, done.data(), done.size());
  }
  if (!ok) c->dead.store(true);
  S->served_reads.fetch_add(1, std::memory_order_relaxed);
  S->served_read_bytes.fetch_add(uint64_t(n), std::memory_order_relaxed);
  return true;
}

// ---------------------------------------------------------------- write serve

// Running data frame for a registered native write session
static bool data_serve_write(DataServer* S, DataConn* c, const uint8_t* frame,
                             uint32_t hlen, uint32_t dlen, uint8_t status,
                             uint64_t req_id, uint32_t seq) {
  std::shared_ptr<WriteSess> ws;
  {
    std::lock_guard<std::mutex> g(c->smu);
    auto it = c->sess.find(req_id);
    if (it == c->sess.end()) return false;
    ws = it->second;
  }
  const uint8_t* h = frame + kMetaProto;
  int64_t off = -1;
  bool positional = hlen && mp_find_int(h, h + hlen, "off", 3, &off);
  uint64_t dst = positional ? uint64_t(off) : ws->pos.load();
  if (dst + dlen > ws->reserved) return false;  // overflow: Python decides
  const uint8_t* payload = frame + kMetaProto + hlen;
  if (dlen) {
    if (ws->kind == 0) {
      Arena* a = get_arena(ws->arena);
      if (a->is_dev()) {
        dev_write(a, ws->aoff + dst, payload, dlen);
      } else {
        std::memcpy((uint8_t*)a->base + ws->aoff + dst, payload, dlen);
      }
    } else {
      size_t left = dlen;
      const uint8_t* p = payload;
      off_t fo = off_t(dst);
      while (left) {
        ssize_t w = pwrite(ws->fd, p, left, fo);
        if (w < 0) {
          if (errno == EINTR) continue;
          // bytes may already be placed: forwarding the frame to Python
          // would re-write it at a stale position.  Fail the stream —
          // the client aborts and re-places the block elsewhere.
          c->dead.store(true);
          return true;
        }
        p += w;
        fo += w;
        left -= size_t(w);
      }
    }
  }
  uint64_t endpos = dst + dlen;
  uint64_t cur = ws->pos.load();
  while (endpos > cur && !ws->pos.compare_exchange_weak(cur, endpos)) {
  }
  S->served_writes.fetch_add(1, std::memory_order_relaxed);
  S->served_write_bytes.fetch_add(dlen, std::memory_order_relaxed);
  // ack (client pipelines a window)
  std::string ack = meta_proto(0, 0, kCodeWriteBlock,
                               uint8_t((2 << 4) | (status & 0xF)), req_id,
                               seq);
  std::lock_guard<std::mutex> wg(c->wmu);
  if (!fd_write_all(c->fd, ack.data(), ack.size())) c->dead.store(true);
  return true;
}

// ---------------------------------------------------------------- loop

static void data_handle_frame(DataServer* S, const std::shared_ptr<DataConn>& c,
                              const uint8_t* frame, uint32_t hlen,
                              uint32_t dlen) {
  uint8_t code = frame[8], status = frame[9];
  uint64_t req_id = rd_u64be(frame + 10);
  uint32_t seq = rd_u32be(frame + 18);
  uint8_t req_status = status & 0xF;
  if (code == kCodeReadBlock && req_status == 1 /*Open*/) {
    bool served = false;
    try {
      served = data_serve_read(S, c.get(), frame, hlen, status, req_id, seq);
    } catch (const std::exception&) {
      // mid-stream failure (e.g. HIP error): the stream is torn — the
      // only safe recovery is dropping the connection
      c->dead.store(true);
      return;
    }
    if (served) return;
  } else if (code == kCodeWriteBlock && req_status == 2 /*Running*/) {
    bool served = false;
    try {
      served =
          data_serve_write(S, c.get(), frame, hlen, dlen, status, req_id, seq);
    } catch (const std::exception&) {
      c->dead.store(true);
      return;
    }
    if (served) return;
  } else if (code == kCodeKeepalive && hlen == 0 && dlen == 0) {
    std::string out = meta_proto(0, 0, code, uint8_t((3 << 4) | req_status),
                                 req_id, seq);
    std::lock_guard<std::mutex> wg(c->wmu);
    if (!fd_write_all(c->fd, out.data(), out.size())) c->dead.store(true);
    return;
  }
  data_forward(S, c, frame, kMetaProto + hlen + dlen);
}

static void data_readable(DataServer* S, const std::shared_ptr<DataConn>& c) {
  for (;;) {
    // figure out how much the current frame still needs; recv straight
    // into the tail of rbuf (single copy off the socket)
    size_t want = 256 << 10;
    if (c->rlen >= kMetaProto) {
      uint32_t hlen = rd_u32be(c->rbuf.data());
      uint32_t dlen = rd_u32be(c->rbuf.data() + 4);
      if (hlen > kMetaMaxLen || dlen > kMetaMaxLen) {
        data_close_conn(S, c);
        return;
      }
      size_t total = kMetaProto + hlen + dlen;
      if (total > c->rlen) want = total - c->rlen;
    }
    c->rbuf.grow(c->rlen + want, c->rlen);
    ssize_t n = recv(c->fd, c->rbuf.data() + c->rlen, want, 0);
    if (n == 0) {
      data_close_conn(S, c);
      return;
    }
    if (n < 0) {
      if (errno == EINTR) continue;
      if (errno == EAGAIN || errno == EWOULDBLOCK) break;
      data_close_conn(S, c);
      return;
    }
    c->rlen += size_t(n);
    // drain complete frames
    size_t off = 0;
    while (c->rlen - off >= kMetaProto) {
      const uint8_t* p = c->rbuf.data() + off;
      uint32_t hlen = rd_u32be(p);
      uint32_t dlen = rd_u32be(p + 4);
      if (hlen > kMetaMaxLen || dlen > kMetaMaxLen) {
        data_close_conn(S, c);
        return;
      }
      size_t total = kMetaProto + hlen + dlen;
      if (c->rlen - off < total) break;
      data_handle_frame(S, c, p, hlen, dlen);
      off += total;
    }
    if (off) {
      std::memmove(c->rbuf.data(), c->rbuf.data() + off, c->rlen - off);
      c->rlen -= off;
    }
    if (c->rlen == 0 && c->rbuf.capacity() > (8u << 20) && !c->rbuf.pinned) {
      // a 16 MiB write frame would otherwise hold its buffer per conn
      // (pinned buffers stay: re-pinning costs ~1 ms and an HBM write
      // session is usually followed by more)
      c->rbuf.release();
    }
    if (c->dead.load()) {
      data_close_conn(S, c);
      return;
    }
    if (size_t(n) < want) break;  // drained the socket
  }
  if (c->dead.load()) {
    data_close_conn(S, c);
    return;
  }
  struct epoll_event ev;
  ev.events = EPOLLIN | EPOLLRDHUP | EPOLLONESHOT;
  ev.data.u64 = c->id;
  if (epoll_ctl(S->epfd, EPOLL_CTL_MOD, c->fd, &ev) != 0)
    data_close_conn(S, c);
}

static void data_accept(DataServer* S) {
  for (;;) {
    int fd = accept4(S->listen_fd, nullptr, nullptr, SOCK_NONBLOCK);
    if (fd < 0) break;
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
    int sz = 4 << 20;
    setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sz, sizeof sz);
    setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &sz, sizeof sz);
    auto c = std::make_shared<DataConn>();
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
    if (epoll_ctl(S->epfd, EPOLL_CTL_ADD, fd, &ev) != 0) data_close_conn(S, c);
  }
  struct epoll_event ev;
  ev.events = EPOLLIN | EPOLLONESHOT;
  ev.data.u64 = 0;
  epoll_ctl(S->epfd, EPOLL_CTL_MOD, S->listen_fd, &ev);
}

static void data_thread(DataServer* S) {
  while (!S->stopping.load()) {
    struct epoll_event ev;
    int n = epoll_wait(S->epfd, &ev, 1, 500);
    if (n <= 0) continue;
    if (ev.data.u64 == 0) {
      data_accept(S);
      continue;
    }
    std::shared_ptr<DataConn> c;
    {
      std::lock_guard<std::mutex> g(S->conns_mu);
      auto it = S->conns.find(ev.data.u64);
      if (it != S->conns.end()) c = it->second;
    }
    if (!c) continue;
    if (ev.events & (EPOLLHUP | EPOLLERR)) {
      data_close_conn(S, c);
      continue;
    }
    data_readable(S, c);
  }
}

// ---------------------------------------------------------------- api

static int64_t data_create(int listen_fd, int nthreads) {
  auto S = std::make_unique<DataServer>();
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
  DataServer* raw = S.get();
  for (int i = 0; i < nthreads; i++) S->threads.emplace_back(data_thread, raw);
  std::lock_guard<std::mutex> g(g_data_mu);
  int64_t sid = g_data_next++;
  g_data[sid] = std::move(S);
  return sid;
}

static void data_stop_srv(int64_t sid) {
  std::unique_ptr<DataServer> S;
  {
    std::lock_guard<std::mutex> g(g_data_mu);
    auto it = g_data.find(sid);
    if (it == g_data.end()) return;
    S = std::move(it->second);
    g_data.erase(it);
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
  for (auto& kv : S->conns) {
    close(kv.second->fd);
    for (auto& sk : kv.second->sess)
      if (sk.second->fd >= 0) close(sk.second->fd);
  }
}

static void data_block_publish(int64_t sid, int64_t block_id, int kind,
                               int arena, uint64_t aoff, int64_t len,
                               const std::string& path,
                               bool direct = false) {
  DataServer* S = data_get(sid);
  auto b = std::make_shared<DataBlock>();
  b->kind = kind;
  b->arena = arena;
  b->aoff = aoff;
  b->len = len;
  b->path = path;
  b->direct = direct;
  std::unique_lock<std::shared_mutex> lk(S->blk_mu);
  S->blocks[block_id] = std::move(b);
}

// returns the number of in-flight native readers; the caller defers the
// layout deallocation until this reaches 0 (new opens fail immediately)
static int data_block_drop(int64_t sid, int64_t block_id) {
  DataServer* S = data_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->blk_mu);
  auto it = S->blocks.find(block_id);
  if (it == S->blocks.end()) return 0;
  std::shared_ptr<DataBlock> b = it->second;
  b->dead.store(true);
  S->blocks.erase(it);
  int refs = b->refs.load();
  if (refs > 0) S->dying[block_id] = std::move(b);
  return refs;
}

// remaining native readers on a dropped block; reaps the record at 0
static int data_block_refs(int64_t sid, int64_t block_id) {
  DataServer* S = data_get(sid);
  std::unique_lock<std::shared_mutex> lk(S->blk_mu);
  auto it = S->dying.find(block_id);
  if (it == S->dying.end()) return 0;
  int refs = it->second->refs.load();
  if (refs == 0) S->dying.erase(it);
  return refs;
}

static void data_write_register(int64_t sid, uint64_t conn_id, uint64_t req_id,
                                int kind, int arena, uint64_t aoff,
                                uint64_t reserved, int fd, uint64_t pos) {
  DataServer* S = data_get(sid);
  std::shared_ptr<DataConn> c;
  {
    std::lock_guard<std::mutex> g(S->conns_mu);
    auto it = S->conns.find(conn_id);
    if (it == S->conns.end()) throw std::runtime_error("conn gone");
    c = it->second;
  }
  auto ws = std::make_shared<WriteSess>();
  ws->kind = kind;
  ws->arena = arena;
  ws->aoff = aoff;
  ws->reserved = reserved;
  ws->fd = fd >= 0 ? dup(fd) : -1;
  ws->pos.store(pos);
  if (kind == 0 && arena >= 0 && get_arena(arena)->is_dev()) {
    // HBM destination: upgrade this conn's receive buffer to pinned
    // pages at its next growth so dev_write DMAs straight from it
    c->rbuf.want_pinned.store(true, std::memory_order_relaxed);
  }
  std::lock_guard<std::mutex> g(c->smu);
  c->sess[req_id] = std::move(ws);
}

// returns the append watermark (finalize default length); -1 if unknown
static int64_t data_write_unregister(int64_t sid, uint64_t conn_id,
                                     uint64_t req_id) {
  DataServer* S = data_get(sid);
  std::shared_ptr<DataConn> c;
  {
    std::lock_guard<std::mutex> g(S->conns_mu);
    auto it = S->conns.find(conn_id);
    if (it == S->conns.end()) return -1;
    c = it->second;
  }
  std::lock_guard<std::mutex> g(c->smu);
  auto it = c->sess.find(req_id);
  if (it == c->sess.end()) return -1;
  int64_t pos = int64_t(it->second->pos.load());
  if (it->second->fd >= 0) close(it->second->fd);
  c->sess.erase(it);
  return pos;
}

static int data_eventfd(int64_t sid) { return data_get(sid)->fq_efd; }

static py::list data_forward_pop(int64_t sid, int timeout_ms, int max_items) {
  DataServer* S = data_get(sid);
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

static bool data_send(int64_t sid, uint64_t conn_id, py::bytes data) {
  DataServer* S = data_get(sid);
  std::shared_ptr<DataConn> c;
  {
    std::lock_guard<std::mutex> g(S->conns_mu);
    auto it = S->conns.find(conn_id);
    if (it == S->conns.end()) return false;
    c = it->second;
  }
  std::string d = data;
  py::gil_scoped_release rel;
  std::lock_guard<std::mutex> wg(c->wmu);
  if (c->dead.load()) return false;
  return fd_write_all(c->fd, d.data(), d.size());
}

static py::dict data_stats(int64_t sid) {
  DataServer* S = data_get(sid);
  py::dict d;
  d["served_reads"] = S->served_reads.load();
  d["served_read_bytes"] = S->served_read_bytes.load();
  d["served_writes"] = S->served_writes.load();
  d["served_write_bytes"] = S->served_write_bytes.load();
  d["forwarded"] = S->forwarded.load();
  d["conns_total"] = S->conns_total.load();
  {
    std::shared_lock<std::shared_mutex> lk(S->blk_mu);
    d["published_blocks"] = S->blocks.size();
  }
  return d;
}

// ============================================================= native client

// Blocking pooled client for the data plane: reads stream straight into the
// caller's buffer (single copy off the socket), writes stream from the
// caller's buffer with a pipelined ack window.  GIL released throughout.

struct DataPool {
  std::mutex mu;
  std::unordered_map<std::string, std::vector<int>> free_fds;
};
static DataPool g_dpool;

static int dc_connect(const std::string& host, int port) {
  struct sockaddr_in sa = {};
  sa.sin_family = AF_INET;
  sa.sin_port = htons(uint16_t(port));
  if (inet_pton(AF_INET, host.c_str(), &sa.sin_addr) != 1) {
    // resolve via getaddrinfo for hostnames
    struct addrinfo hints = {}, *res = nullptr;
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    if (getaddrinfo(host.c_str(), nullptr, &hints, &res) != 0 || !res)
      return -1;
    sa.sin_addr = ((struct sockaddr_in*)res->ai_addr)->sin_addr;
    freeaddrinfo(res);
  }
  int fd = socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) return -1;
  if (connect(fd, (struct sockaddr*)&sa, sizeof sa) != 0) {
    close(fd);
    return -1;
  }
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
  int sz = 4 << 20;
  setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sz, sizeof sz);
  setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &sz, sizeof sz);
  return fd;
}

static int dc_acquire(const std::string& host, int port) {
  std::string key = host + ":" + std::to_string(port);
  {
    std::lock_guard<std::mutex> g(g_dpool.mu);
    auto& v = g_dpool.free_fds[key];
    if (!v.empty()) {
      int fd = v.back();
      v.pop_back();
      return fd;
    }
  }
  return dc_connect(host, port);
}

static void dc_release(const std::string& host, int port, int fd, bool ok) {
  if (!ok) {
    close(fd);
    return;
  }
  std::string key = host + ":" + std::to_string(port);
  std::lock_guard<std::mutex> g(g_dpool.mu);
  auto& v = g_dpool.free_fds[key];
  if (v.size() >= 16) {
    close(fd);
    return;
  }
  v.push_back(fd);
}

static bool dc_read_exact(int fd, uint8_t* p, size_t n) {
  while (n) {
    ssize_t r = recv(fd, p, n, 0);
    if (r > 0) { p += r; n -= size_t(r); continue; }
    if (r < 0 && errno == EINTR) continue;
    return false;
  }
  return true;
}

static std::atomic<uint64_t> g_dc_req{1u << 20};

// returns (resp_status, final_header_bytes, bytes_read)
static py::tuple data_read_into(const std::string& host, int port,
                                int64_t block_id, int64_t offset,
                                int64_t length, py::buffer dst,
                                uint64_t dst_off, int64_t chunk) {
  py::buffer_info info = dst.request(true);
  uint64_t cap = (uint64_t)info.size * (uint64_t)info.itemsize;
  uint8_t* out = (uint8_t*)info.ptr + dst_off;
  if (length >= 0 && dst_off + uint64_t(length) > cap)
    throw std::runtime_error("dst buffer too small");
  std::string err_hdr;
  uint8_t final_status = 5;
  int64_t got = 0;
  {
    py::gil_scoped_release rel;
    uint64_t req = g_dc_req.fetch_add(1);
    // Open frame
    std::string h;
    h.push_back(char(0x84));
    mp_str(h, "block_id", 8);
    mp_uint(h, uint64_t(block_id));
    mp_str(h, "offset", 6);
    mp_uint(h, uint64_t(offset));
    mp_str(h, "length", 6);
    mp_uint(h, uint64_t(length));
    mp_str(h, "chunk_size", 10);
    mp_uint(h, uint64_t(chunk));
    std::string head = meta_proto(uint32_t(h.size()), 0, kCodeReadBlock,
                                  uint8_t(1) /*Open req*/, req, 0);
    head += h;
    // a pooled fd can be stale (server restarted, port reused): retry
    // ONCE on a fresh socket if nothing was consumed yet
    for (int attempt = 0; attempt < 2; attempt++) {
      int fd = attempt == 0 ? dc_acquire(host, port) : dc_connect(host, port);
      if (fd < 0) {
        if (attempt == 0) continue;
        throw std::runtime_error("data connect failed");
      }
      bool keep = false;
      bool progressed = false;
      got = 0;
      if (fd_write_all(fd, head.data(), head.size())) {
        // consume frames until Complete/Error
        std::vector<uint8_t> hdr_buf;
        for (;;) {
          uint8_t proto[kMetaProto];
          if (!dc_read_exact(fd, proto, kMetaProto)) break;
          progressed = true;
          uint32_t hlen = rd_u32be(proto);
          uint32_t dlen = rd_u32be(proto + 4);
          if (hlen > kMetaMaxLen || dlen > kMetaMaxLen) break;
          uint8_t st = proto[9] >> 4;
          hdr_buf.resize(hlen);
          if (hlen && !dc_read_exact(fd, hdr_buf.data(), hlen)) break;
          if (dlen) {
            if (got + dlen > int64_t(cap - dst_off)) break;  // overflow
            if (!dc_read_exact(fd, out + got, dlen)) break;
            got += dlen;
          }
          if (st == 3 || st == 5) {  // Complete / Error
            final_status = st;
            err_hdr.assign((const char*)hdr_buf.data(), hlen);
            keep = true;
            break;
          }
        }
      }
      dc_release(host, port, fd, keep);
      if (keep || progressed) break;  // real outcome (or torn mid-stream)
    }
  }
  return py::make_tuple(int(final_status), py::bytes(err_hdr), got);
}

// ------------------------------------------------ streaming write session
//
// The client-side analog of the reference's BlockWriter remote stream
// (block_writer.rs open/write/commit over one connection), kept in C++
// so FsWriter's executor threads stream chunks GIL-free: dw_open does
// the Open round-trip on a pooled conn, dw_write sends one windowed
// chunk frame, dw_commit completes (or dw_abort cancels) and returns
// the final header.

struct DwSession {
  std::string host;
  int port = 0;
  int fd = -1;
  uint64_t req = 0;
  uint32_t seq = 0;
  int inflight = 0;
  int window = 8;
  bool failed = false;
};

static std::mutex g_dw_mu;
static std::unordered_map<int64_t, DwSession> g_dw;
static int64_t g_dw_next = 1;

static bool dw_read_reply(int fd, uint8_t* st_out, std::string* hdr_out) {
  uint8_t proto[kMetaProto];
  if (!dc_read_exact(fd, proto, kMetaProto)) return false;
  uint32_t hlen = rd_u32be(proto);
  uint32_t dlen = rd_u32be(proto + 4);
  if (hlen > kMetaMaxLen || dlen > kMetaMaxLen) return false;
  std::vector<uint8_t> tmp(hlen + dlen);
  if ((hlen + dlen) && !dc_read_exact(fd, tmp.data(), hlen + dlen))
    return false;
  *st_out = proto[9] >> 4;
  if (hdr_out) hdr_out->assign((const char*)tmp.data(), hlen);
  return true;
}

// (handle>0, status, header) — handle 0 on connect failure or rejection
static py::tuple dw_open(const std::string& host, int port, int64_t block_id,
                         int64_t reserve, const std::string& tier,
                         bool reopen, int window) {
  std::string hdr;
  uint8_t st = 5;
  int fd = -1;
  uint64_t req = 0;
  {
    py::gil_scoped_release rel;
    req = g_dc_req.fetch_add(1);
    std::string h;
    h.push_back(char(0x84));
    mp_str(h, "block_id", 8);
    mp_uint(h, uint64_t(block_id));
    mp_str(h, "reserve", 7);
    mp_uint(h, uint64_t(reserve));
    mp_str(h, "tier", 4);
    mp_str(h, tier);
    mp_str(h, "reopen", 6);
    h.push_back(char(reopen ? 0xc3 : 0xc2));
    std::string head = meta_proto(uint32_t(h.size()), 0, kCodeWriteBlock,
                                  uint8_t(1), req, 0);
    head += h;
    for (int attempt = 0; attempt < 2; attempt++) {
      fd = attempt == 0 ? dc_acquire(host, port) : dc_connect(host, port);
      if (fd < 0) continue;
      if (fd_write_all(fd, head.data(), head.size()) &&
          dw_read_reply(fd, &st, &hdr))
        break;
      close(fd);
      fd = -1;
    }
  }
  if (fd < 0) return py::make_tuple(int64_t(0), 5, py::bytes());
  if (st == 5) {  // rejected: the stream is clean, keep the conn pooled
    dc_release(host, port, fd, true);
    return py::make_tuple(int64_t(0), 5, py::bytes(hdr));
  }
  std::lock_guard<std::mutex> g(g_dw_mu);
  int64_t id = g_dw_next++;
  DwSession& ses = g_dw[id];
  ses.host = host;
  ses.port = port;
  ses.fd = fd;
  ses.req = req;
  ses.seq = 1;
  ses.window = window > 0 ? window : 8;
  return py::make_tuple(id, int(st), py::bytes(hdr));
}

static DwSession* dw_get(int64_t id) {
  std::lock_guard<std::mutex> g(g_dw_mu);
  auto it = g_dw.find(id);
  return it == g_dw.end() ? nullptr : &it->second;
}

// send one data frame (chunked internally), draining acks past the
// window.  Returns false when the stream failed (caller aborts).
static bool dw_write(int64_t id, py::buffer src, uint64_t src_off,
                     int64_t n, int64_t chunk) {
  DwSession* ses = dw_get(id);
  if (!ses || ses->failed) return false;
  py::buffer_info info = src.request(false);
  uint64_t cap = (uint64_t)info.size * (uint64_t)info.itemsize;
  if (src_off + uint64_t(n) > cap)
    throw std::runtime_error("src buffer too small");
  const uint8_t* in = (const uint8_t*)info.ptr + src_off;
  bool ok = true;
  {
    py::gil_scoped_release rel;
    int64_t pos = 0;
    if (chunk <= 0) chunk = 4 << 20;
    while (pos < n && ok) {
      int64_t cn = std::min<int64_t>(chunk, n - pos);
      std::string ph = meta_proto(0, uint32_t(cn), kCodeWriteBlock,
                                  uint8_t(2), ses->req, ses->seq++);
      struct iovec iov[2] = {{(void*)ph.data(), ph.size()},
                             {(void*)(in + pos), size_t(cn)}};
      struct msghdr mh = {};
      mh.msg_iov = iov;
      mh.msg_iovlen = 2;
      size_t sent = 0, total = ph.size() + size_t(cn);
      while (sent < total && ok) {
        ssize_t w = sendmsg(ses->fd, &mh, MSG_NOSIGNAL);
        if (w < 0) {
          if (errno == EINTR) continue;
          ok = false;
          break;
        }
        sent += size_t(w);
        size_t adv = size_t(w);
        for (int i = 0; i < 2 && adv; i++) {
          size_t take = std::min(adv, iov[i].iov_len);
          iov[i].iov_base = (char*)iov[i].iov_base + take;
          iov[i].iov_len -= take;
          adv -= take;
        }
        while (mh.msg_iovlen && mh.msg_iov->iov_len == 0) {
          mh.msg_iov++;
          mh.msg_iovlen--;
        }
      }
      pos += cn;
      ses->inflight++;
      while (ok && ses->inflight >= ses->window) {
        uint8_t st = 0;
        ok = dw_read_reply(ses->fd, &st, nullptr);
        ses->inflight--;
        if (st == 5) ok = false;
      }
    }
  }
  if (!ok) ses->failed = true;
  return ok;
}

// read acks until nothing is in flight: callers that are about to
// rewrite bytes on a SEPARATE reopen stream must know the appends have
// been consumed by the server first (two connections are unordered)
static bool dw_drain(int64_t id) {
  DwSession* ses = dw_get(id);
  if (!ses || ses->failed) return false;
  bool ok = true;
  {
    py::gil_scoped_release rel;
    while (ok && ses->inflight > 0) {
      uint8_t st = 0;
      ok = dw_read_reply(ses->fd, &st, nullptr);
      ses->inflight--;
      if (st == 5) ok = false;
    }
  }
  if (!ok) ses->failed = true;
  return ok;
}

static void dw_close(int64_t id, bool keep) {
  std::lock_guard<std::mutex> g(g_dw_mu);
  auto it = g_dw.find(id);
  if (it == g_dw.end()) return;
  dc_release(it->second.host, it->second.port, it->second.fd, keep);
  g_dw.erase(it);
}

// (status, final_header): drain acks, send Complete (or no_finalize
// close), read the final reply
static py::tuple dw_commit(int64_t id, int64_t length, bool no_finalize) {
  DwSession* ses = dw_get(id);
  if (!ses) return py::make_tuple(5, py::bytes());
  std::string hdr;
  uint8_t st = 5;
  bool ok = !ses->failed;
  {
    py::gil_scoped_release rel;
    while (ok && ses->inflight > 0) {
      uint8_t ast = 0;
      ok = dw_read_reply(ses->fd, &ast, nullptr);
      ses->inflight--;
      if (ast == 5) ok = false;
    }
    if (ok) {
      std::string fh;
      fh.push_back(char(0x81));
      if (no_finalize) {
        mp_str(fh, "no_finalize", 11);
        fh.push_back(char(0xc3));
      } else {
        mp_str(fh, "length", 6);
        mp_uint(fh, uint64_t(length));
      }
      std::string cf = meta_proto(uint32_t(fh.size()), 0, kCodeWriteBlock,
                                  uint8_t(3), ses->req, ses->seq++);
      cf += fh;
      ok = fd_write_all(ses->fd, cf.data(), cf.size());
      if (ok) ok = dw_read_reply(ses->fd, &st, &hdr);
    }
  }
  dw_close(id, ok && st != 5);
  return py::make_tuple(ok ? int(st) : 5, py::bytes(hdr));
}

static void dw_abort(int64_t id) {
  DwSession* ses = dw_get(id);
  if (!ses) return;
  {
    py::gil_scoped_release rel;
    std::string cf = meta_proto(0, 0, kCodeWriteBlock, uint8_t(4) /*Cancel*/,
                                ses->req, ses->seq++);
    (void)fd_write_all(ses->fd, cf.data(), cf.size());
  }
  dw_close(id, false);
}

// returns (resp_status, final_header_bytes)
static py::tuple data_write_from(const std::string& host, int port,
                                 int64_t block_id, int64_t reserve,
                                 const std::string& tier, py::buffer src,
                                 uint64_t src_off, int64_t length,
                                 int64_t chunk, int window, bool reopen,
                                 int64_t finalize_len) {
  py::buffer_info info = src.request(false);
  uint64_t cap = (uint64_t)info.size * (uint64_t)info.itemsize;
  if (src_off + uint64_t(length) > cap)
    throw std::runtime_error("src buffer too small");
  const uint8_t* in = (const uint8_t*)info.ptr + src_off;
  std::string fin_hdr;
  uint8_t final_status = 5;
  {
    py::gil_scoped_release rel;
    uint64_t req = g_dc_req.fetch_add(1);
    uint32_t seq = 0;
    std::string h;
    h.push_back(char(0x84));
    mp_str(h, "block_id", 8);
    mp_uint(h, uint64_t(block_id));
    mp_str(h, "reserve", 7);
    mp_uint(h, uint64_t(reserve));
    mp_str(h, "tier", 4);
    mp_str(h, tier);
    mp_str(h, "reopen", 6);
    h.push_back(char(reopen ? 0xc3 : 0xc2));
    std::string head = meta_proto(uint32_t(h.size()), 0, kCodeWriteBlock,
                                  uint8_t(1), req, seq++);
    head += h;

    auto read_reply = [&](int fd, uint8_t* st_out,
                          std::string* hdr_out) -> bool {
      uint8_t proto[kMetaProto];
      if (!dc_read_exact(fd, proto, kMetaProto)) return false;
      uint32_t hlen = rd_u32be(proto);
      uint32_t dlen = rd_u32be(proto + 4);
      if (hlen > kMetaMaxLen || dlen > kMetaMaxLen) return false;
      std::vector<uint8_t> tmp(hlen + dlen);
      if ((hlen + dlen) && !dc_read_exact(fd, tmp.data(), hlen + dlen))
        return false;
      *st_out = proto[9] >> 4;
      if (hdr_out) hdr_out->assign((const char*)tmp.data(), hlen);
      return true;
    };

    // Open round-trip; a pooled fd can be stale (server restart, port
    // reuse) so retry ONCE on a fresh socket before giving up
    int fd = -1;
    uint8_t st = 0;
    for (int attempt = 0; attempt < 2; attempt++) {
      fd = attempt == 0 ? dc_acquire(host, port) : dc_connect(host, port);
      if (fd < 0) continue;
      if (fd_write_all(fd, head.data(), head.size()) &&
          read_reply(fd, &st, &fin_hdr))
        break;
      close(fd);
      fd = -1;
    }
    if (fd < 0) throw std::runtime_error("data write open failed");
    if (st == 5) {  // Open rejected (e.g. BlockInWriting): clean stream
      final_status = 5;
      dc_release(host, port, fd, true);
    } else {
      // stream chunks with a pipelined ack window
      bool ok = true;
      bool keep = false;
      int inflight = 0;
      int64_t pos = 0;
      while (pos < length && ok) {
        int64_t cn = std::min<int64_t>(chunk, length - pos);
        std::string ph = meta_proto(0, uint32_t(cn), kCodeWriteBlock,
                                    uint8_t(2), req, seq++);
        struct iovec iov[2] = {{(void*)ph.data(), ph.size()},
                               {(void*)(in + pos), size_t(cn)}};
        struct msghdr mh = {};
        mh.msg_iov = iov;
        mh.msg_iovlen = 2;
        size_t sent = 0, total = ph.size() + size_t(cn);
        while (sent < total && ok) {
          ssize_t w = sendmsg(fd, &mh, MSG_NOSIGNAL);
          if (w < 0) {
            if (errno == EINTR) continue;
            ok = false;
            break;
          }
          sent += size_t(w);
          size_t adv = size_t(w);
          for (int i = 0; i < 2 && adv; i++) {
            size_t take = std::min(adv, iov[i].iov_len);
            iov[i].iov_base = (char*)iov[i].iov_base + take;
            iov[i].iov_len -= take;
            adv -= take;
          }
          while (mh.msg_iovlen && mh.msg_iov->iov_len == 0) {
            mh.msg_iov++;
            mh.msg_iovlen--;
          }
        }
        pos += cn;
        inflight++;
        while (ok && inflight >= window) {
          ok = read_reply(fd, &st, nullptr);
          inflight--;
          if (st == 5) ok = false;
        }
      }
      while (ok && inflight > 0) {
        ok = read_reply(fd, &st, nullptr);
        inflight--;
        if (st == 5) ok = false;
      }
      if (ok) {
        std::string fh;
        fh.push_back(char(0x81));
        mp_str(fh, "length", 6);
        mp_uint(fh, uint64_t(finalize_len >= 0 ? finalize_len : length));
        std::string cf = meta_proto(uint32_t(fh.size()), 0, kCodeWriteBlock,
                                    uint8_t(3) /*Complete*/, req, seq++);
        cf += fh;
        ok = fd_write_all(fd, cf.data(), cf.size());
        if (ok) ok = read_reply(fd, &st, &fin_hdr);
        if (ok) {
          final_status = st;
          keep = true;
        }
      }
      dc_release(host, port, fd, keep);
    }
  }
  return py::make_tuple(int(final_status), py::bytes(fin_hdr));
}
