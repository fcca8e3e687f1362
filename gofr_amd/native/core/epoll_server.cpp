// gofr_amd native ingress: multi-reactor epoll HTTP/1.1 listener staging
// request bytes for the GPU batch engine.
//
// Role of the reference's net/http accept loop (pkg/gofr/httpServer.go:
// 24-36, goroutine per connection) re-designed for the MI355X data plane:
// instead of a handler per connection, N reactor threads (each its own
// SO_REUSEPORT listener + epoll set — the kernel load-balances accepts)
// read COMPLETE HTTP requests (headers + content-length body) into
// per-connection buffers; harvest() batch-copies them straight into the
// engine's pinned ingress ring (no Python in the byte path). Responses
// are written back from the engine's pinned egress buffer, honoring
// keep-alive. A single reactor measured ~1.45M req/s on MI355X-class
// hosts — the multi-reactor split removes that ceiling.
//
// Python surface (pybind11 module gofr_amd._core):
//   s = EpollServer(port, max_req, threads)
//   s.start()
//   n, nbytes = s.harvest(buf_ptr, buf_cap, off_ptr, len_ptr, conn_ptr,
//                         max_n, window_us)      # blocks <= window_us
//   s.send(conn_ptr, n, out_ptr, roff_ptr, rlen_ptr)
//   s.stop()

#include <pybind11/pybind11.h>

#include <arpa/inet.h>
#include <atomic>
#include <cerrno>
#include <cstdlib>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <fcntl.h>
#include <functional>
#include <memory>
#include <mutex>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <string>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <thread>
#include <unistd.h>
#include <unordered_map>
#include <vector>

#include "hpack_tables.h"

namespace py = pybind11;

namespace {

// ---------------------------------------------------------------------------
// HTTP/2 + HPACK (server-side): the gRPC ingress lives in the SAME
// epoll reactors as HTTP/1.1 (VERDICT r1 item 7 — the reference serves
// gRPC on its production listener, pkg/gofr/grpc.go:32-47). A
// connection that opens with the h2c client preface switches to frame
// mode; unary gRPC request messages flow into the batched GPU codec
// via harvest_grpc, responses go back as precomputed
// HEADERS/DATA/trailers frame sequences.
// ---------------------------------------------------------------------------

static const char H2_PREFACE[] = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";
#define H2_PREFACE_LEN 24

// huffman decode tree built once from HUFF_TBL
struct HuffNode { int16_t next[2] = {-1, -1}; int16_t sym = -1; };
static std::vector<HuffNode> g_huff;
static std::mutex g_huff_mu;

static void huff_init() {
    std::lock_guard<std::mutex> lk(g_huff_mu);
    if (!g_huff.empty()) return;
    std::vector<HuffNode> t(1);
    for (int s = 0; s < 257; ++s) {
        uint32_t code = HUFF_TBL[s].code;
        int len = HUFF_TBL[s].len;
        int node = 0;
        for (int b = len - 1; b >= 0; --b) {
            const int bit = (code >> b) & 1;
            if (t[node].next[bit] < 0) {
                t[node].next[bit] = (int16_t)t.size();
                t.emplace_back();
            }
            node = t[node].next[bit];
        }
        t[node].sym = (int16_t)s;
    }
    g_huff.swap(t);
}

static bool huff_decode(const uint8_t* p, size_t n, std::string* out) {
    int node = 0;
    for (size_t i = 0; i < n; ++i) {
        for (int b = 7; b >= 0; --b) {
            node = g_huff[node].next[(p[i] >> b) & 1];
            if (node < 0) return false;
            if (g_huff[node].sym >= 0) {
                if (g_huff[node].sym == 256) return false;  // EOS in data
                out->push_back((char)g_huff[node].sym);
                node = 0;
            }
        }
    }
    return true;  // trailing bits are EOS-prefix padding (unchecked ok)
}

struct HpackDec {
    std::deque<std::pair<std::string, std::string>> dyn;  // front=newest
    size_t dyn_bytes = 0, max_bytes = 4096;

    bool entry(size_t idx, std::string* n, std::string* v) {
        if (idx >= 1 && idx <= 61) {
            *n = HPACK_STATIC[idx - 1][0];
            *v = HPACK_STATIC[idx - 1][1];
            return true;
        }
        const size_t d = idx - 62;
        if (d >= dyn.size()) return false;
        *n = dyn[d].first;
        *v = dyn[d].second;
        return true;
    }

    void add(const std::string& n, const std::string& v) {
        dyn_bytes += n.size() + v.size() + 32;
        dyn.emplace_front(n, v);
        while (dyn_bytes > max_bytes && !dyn.empty()) {
            dyn_bytes -= dyn.back().first.size() +
                         dyn.back().second.size() + 32;
            dyn.pop_back();
        }
    }

    static bool read_int(const uint8_t* p, size_t n, size_t* pos,
                         int prefix, uint64_t* out) {
        if (*pos >= n) return false;
        const uint64_t mask = (1u << prefix) - 1;
        uint64_t v = p[(*pos)++] & mask;
        if (v == mask) {
            int shift = 0;
            while (true) {
                if (*pos >= n || shift > 56) return false;
                const uint8_t b = p[(*pos)++];
                v += (uint64_t)(b & 0x7F) << shift;
                if (!(b & 0x80)) break;
                shift += 7;
            }
        }
        *out = v;
        return true;
    }

    bool read_str(const uint8_t* p, size_t n, size_t* pos,
                  std::string* out) {
        if (*pos >= n) return false;
        const bool huff = (p[*pos] & 0x80) != 0;
        uint64_t len;
        if (!read_int(p, n, pos, 7, &len)) return false;
        if (*pos + len > n) return false;
        if (huff) {
            if (!huff_decode(p + *pos, (size_t)len, out)) return false;
        } else {
            out->assign((const char*)p + *pos, (size_t)len);
        }
        *pos += (size_t)len;
        return true;
    }

    bool decode(const uint8_t* p, size_t n,
                std::vector<std::pair<std::string, std::string>>* out) {
        size_t pos = 0;
        while (pos < n) {
            const uint8_t b = p[pos];
            std::string name, val;
            if (b & 0x80) {  // indexed
                uint64_t idx;
                if (!read_int(p, n, &pos, 7, &idx) ||
                    !entry((size_t)idx, &name, &val))
                    return false;
                out->emplace_back(name, val);
            } else if (b & 0x40) {  // literal with incremental indexing
                uint64_t idx;
                if (!read_int(p, n, &pos, 6, &idx)) return false;
                if (idx) {
                    std::string dummy;
                    if (!entry((size_t)idx, &name, &dummy)) return false;
                } else if (!read_str(p, n, &pos, &name)) {
                    return false;
                }
                if (!read_str(p, n, &pos, &val)) return false;
                add(name, val);
                out->emplace_back(name, val);
            } else if ((b & 0xE0) == 0x20) {  // table size update
                uint64_t sz;
                if (!read_int(p, n, &pos, 5, &sz)) return false;
                max_bytes = (size_t)sz;
                while (dyn_bytes > max_bytes && !dyn.empty()) {
                    dyn_bytes -= dyn.back().first.size() +
                                 dyn.back().second.size() + 32;
                    dyn.pop_back();
                }
            } else {  // literal without indexing / never indexed
                uint64_t idx;
                if (!read_int(p, n, &pos, 4, &idx)) return false;
                if (idx) {
                    std::string dummy;
                    if (!entry((size_t)idx, &name, &dummy)) return false;
                } else if (!read_str(p, n, &pos, &name)) {
                    return false;
                }
                if (!read_str(p, n, &pos, &val)) return false;
                out->emplace_back(name, val);
            }
        }
        return true;
    }
};

struct H2Stream {
    int path_id = -2;       // -2 unknown path, >=0 registered
    std::string data;       // DATA payload accumulation
    std::string hdr_frag;   // CONTINUATION accumulation
    bool headers_done = false;
    bool end_stream_pending = false;  // END_STREAM rode the HEADERS
};

struct H2State {
    HpackDec dec;
    std::unordered_map<uint32_t, H2Stream> streams;
    uint32_t cont_sid = 0;  // stream awaiting CONTINUATION (0 = none)
};

struct Conn {
    int fd = -1;
    std::string rbuf;      // bytes read, not yet parsed into a request
    std::string wbuf;      // bytes pending write
    bool close_after_write = false;
    bool dead = false;
    uint64_t id = 0;
    int mode = 0;          // 0 undetected, 1 HTTP/1.1, 2 h2c
    std::unique_ptr<H2State> h2;
};

struct PendingReq {
    uint64_t conn_id;
    std::string bytes;
};

struct PendingGrpc {
    uint64_t conn_id;
    uint32_t stream_id;
    int path_id;           // -2 bad frame, -1 unregistered, >=0 fast
    std::string msg;       // protobuf message bytes (prefix stripped)
};

// conn ids: [63:48] reactor index, [47:0] per-reactor serial
static inline int id_reactor(uint64_t id) { return (int)(id >> 48); }

// RFC 9112 §7.1 chunked framing: length of the chunked body starting
// at `start` through the end of the terminating blank line, or -1 if
// more bytes are needed, -2 if malformed.
static long chunked_body_len(const std::string& b, size_t start) {
    size_t pos = start;
    while (true) {
        const size_t eol = b.find("\r\n", pos);
        if (eol == std::string::npos)
            return b.size() - pos > 18 ? -2 : -1;  // size line too long
        char* end = nullptr;
        const std::string tok = b.substr(pos, eol - pos);
        const long size = strtol(tok.c_str(), &end, 16);
        if (end == tok.c_str() || size < 0) return -2;
        pos = eol + 2;
        if (size == 0) {
            // trailer section: lines until the blank line
            while (true) {
                const size_t e2 = b.find("\r\n", pos);
                if (e2 == std::string::npos) return -1;
                if (e2 == pos) return (long)(pos + 2 - start);
                pos = e2 + 2;
            }
        }
        if (pos + (size_t)size + 2 > b.size()) return -1;
        pos += (size_t)size + 2;
    }
}

class Reactor {
public:
    Reactor(int idx, int port, size_t max_req)
        : idx_(idx), port_(port), max_req_(max_req) {}

    ~Reactor() { stop(); }

    int bind_and_listen() {
        listen_fd_ = ::socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK, 0);
        if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
        int one = 1;
        setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
        setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEPORT, &one, sizeof(one));
        sockaddr_in addr{};
        addr.sin_family = AF_INET;
        addr.sin_addr.s_addr = INADDR_ANY;
        addr.sin_port = htons((uint16_t)port_);
        if (bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0)
            throw std::runtime_error("bind() failed: " +
                                     std::string(strerror(errno)));
        if (port_ == 0) {
            socklen_t alen = sizeof(addr);
            getsockname(listen_fd_, (sockaddr*)&addr, &alen);
            port_ = ntohs(addr.sin_port);
        }
        if (listen(listen_fd_, 4096) != 0)
            throw std::runtime_error("listen() failed");
        ep_ = epoll_create1(0);
        epoll_event ev{};
        ev.events = EPOLLIN;
        ev.data.u64 = 0;  // listener marker
        epoll_ctl(ep_, EPOLL_CTL_ADD, listen_fd_, &ev);
        return port_;
    }

    void run() {
        running_ = true;
        loop_ = std::thread([this] { event_loop(); });
    }

    void stop() {
        if (!running_) return;
        running_ = false;
        if (loop_.joinable()) loop_.join();
        for (auto& kv : conns_) ::close(kv.second->fd);
        conns_.clear();
        if (listen_fd_ >= 0) ::close(listen_fd_);
        if (ep_ >= 0) ::close(ep_);
        listen_fd_ = ep_ = -1;
    }

    void queue_writes(const uint64_t* cids, const int* rows, int nrows,
                      const uint8_t* out, const int32_t* roffs,
                      const int32_t* rlens) {
        std::lock_guard<std::recursive_mutex> lk(wmu_);
        for (int k = 0; k < nrows; ++k) {
            const int i = rows[k];
            auto it = conn_index_.find(cids[i]);
            if (it == conn_index_.end()) continue;
            const char* resp = (const char*)out + roffs[i];
            const size_t rl = (size_t)rlens[i];
            it->second->wbuf.append(resp, rl);
            // honor the engine's Connection: close verdict (headers
            // precede the body; cap the scan)
            const size_t scan = rl < 400 ? rl : 400;
            if (memmem(resp, scan, "Connection: close", 17) != nullptr)
                it->second->close_after_write = true;
            pending_writes_.push_back(cids[i]);
        }
    }

    long ready_count() {
        std::lock_guard<std::mutex> lk(mu_);
        return (long)ready_.size();
    }

    // pop one ready request (sharded harvest path)
    bool pop_ready(PendingReq* out) {
        std::lock_guard<std::mutex> lk(mu_);
        if (ready_.empty()) return false;
        *out = std::move(ready_.front());
        ready_.pop_front();
        return true;
    }

    // pop the next ready request only if it fits in `room` bytes
    bool pop_ready_fit(PendingReq* out, long room) {
        std::lock_guard<std::mutex> lk(mu_);
        if (ready_.empty() ||
            (long)ready_.front().bytes.size() > room)
            return false;
        *out = std::move(ready_.front());
        ready_.pop_front();
        return true;
    }

    // pop up to `quota` ready requests fitting in `room` bytes under
    // ONE lock acquisition (the per-request lock was the harvest cost)
    int pop_ready_batch(std::vector<PendingReq>* out, int quota,
                        long room) {
        std::lock_guard<std::mutex> lk(mu_);
        int k = 0;
        while (k < quota && !ready_.empty()) {
            const long sz = (long)ready_.front().bytes.size();
            if (sz > room) break;
            room -= sz;
            out->push_back(std::move(ready_.front()));
            ready_.pop_front();
            ++k;
        }
        return k;
    }

    int port() const { return port_; }

private:
    void event_loop() {
        std::vector<epoll_event> events(1024);
        std::string tmp(1 << 16, '\0');
        while (running_) {
            drain_pending_writes();
            const int k = epoll_wait(ep_, events.data(),
                                     (int)events.size(), 1);
            for (int i = 0; i < k; ++i) {
                const uint64_t id = events[i].data.u64;
                if (id == 0) {
                    accept_new();
                    continue;
                }
                auto it = conns_.find(id);
                if (it == conns_.end()) continue;
                Conn& c = *it->second;
                if (events[i].events & (EPOLLHUP | EPOLLERR)) {
                    close_conn(c);
                    continue;
                }
                if (events[i].events & EPOLLIN) {
                    if (!read_ready(c, tmp)) {
                        close_conn(c);
                        continue;
                    }
                }
                if (events[i].events & EPOLLOUT) flush_conn(c);
            }
            gc_dead();
        }
    }

    void accept_new() {
        while (true) {
            const int fd = accept4(listen_fd_, nullptr, nullptr,
                                   SOCK_NONBLOCK);
            if (fd < 0) return;
            int one = 1;
            setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
            const uint64_t id = ((uint64_t)idx_ << 48) | next_id_++;
            auto up = std::make_unique<Conn>();
            Conn& c = *up;
            conns_[id] = std::move(up);
            c.fd = fd;
            c.id = id;
            {
                std::lock_guard<std::recursive_mutex> lk(wmu_);
                conn_index_[id] = &c;
            }
            epoll_event ev{};
            ev.events = EPOLLIN;
            ev.data.u64 = id;
            epoll_ctl(ep_, EPOLL_CTL_ADD, fd, &ev);
        }
    }

    bool read_ready(Conn& c, std::string& tmp) {
        while (true) {
            const ssize_t r = recv(c.fd, tmp.data(), tmp.size(), 0);
            if (r == 0) return false;
            if (r < 0) {
                if (errno == EAGAIN || errno == EWOULDBLOCK) break;
                return false;
            }
            c.rbuf.append(tmp.data(), (size_t)r);
            if (c.rbuf.size() > max_req_) return false;
        }
        // protocol detection: the h2c client preface switches this
        // connection into HTTP/2 frame mode (gRPC ingress)
        if (c.mode == 0) {
            const size_t have = c.rbuf.size();
            const size_t cmp = have < H2_PREFACE_LEN ? have
                                                     : H2_PREFACE_LEN;
            if (memcmp(c.rbuf.data(), H2_PREFACE, cmp) != 0) {
                c.mode = 1;
            } else if (have >= H2_PREFACE_LEN) {
                c.mode = 2;
                c.rbuf.erase(0, H2_PREFACE_LEN);
                huff_init();
                c.h2 = std::make_unique<H2State>();
                std::lock_guard<std::recursive_mutex> lk(wmu_);
                append_frame(&c.wbuf, 0x4, 0, 0, nullptr, 0);  // SETTINGS
                // open the client's connection send window wide
                const uint8_t wu[4] = {0x3f, 0xff, 0xff, 0xff};
                append_frame(&c.wbuf, 0x8, 0, 0, (const char*)wu, 4);
                pending_writes_.push_back(c.id);
            } else {
                return true;  // need more bytes to decide
            }
        }
        if (c.mode == 2) return process_h2(c);
        // slice out complete requests
        while (true) {
            const size_t he = c.rbuf.find("\r\n\r\n");
            if (he == std::string::npos) break;
            size_t clen = 0;
            bool chunked = false;
            // scan headers for content-length / transfer-encoding
            for (size_t p = 0; p < he;) {
                size_t eol = c.rbuf.find("\r\n", p);
                if (eol == std::string::npos || eol > he) eol = he;
                if (eol - p > 15) {
                    static const char k[] = "content-length:";
                    bool match = true;
                    for (int j = 0; j < 15; ++j) {
                        char ch = c.rbuf[p + j];
                        if (ch >= 'A' && ch <= 'Z') ch |= 0x20;
                        if (ch != k[j]) { match = false; break; }
                    }
                    if (match)
                        clen = strtoul(c.rbuf.c_str() + p + 15, nullptr,
                                       10);
                }
                if (eol - p > 18) {
                    static const char t[] = "transfer-encoding:";
                    bool match = true;
                    for (int j = 0; j < 18; ++j) {
                        char ch = c.rbuf[p + j];
                        if (ch >= 'A' && ch <= 'Z') ch |= 0x20;
                        if (ch != t[j]) { match = false; break; }
                    }
                    if (match) {
                        std::string v = c.rbuf.substr(p + 18, eol - p - 18);
                        for (auto& ch : v)
                            if (ch >= 'A' && ch <= 'Z') ch |= 0x20;
                        if (v.find("chunked") != std::string::npos)
                            chunked = true;
                    }
                }
                p = eol + 2;
            }
            size_t total;
            if (chunked) {
                const long blen = chunked_body_len(c.rbuf, he + 4);
                if (blen == -2) return false;  // malformed framing
                if (blen < 0) break;           // need more bytes
                total = he + 4 + (size_t)blen;
            } else {
                total = he + 4 + clen;
            }
            if (c.rbuf.size() < total) break;
            PendingReq req;
            req.conn_id = c.id;
            req.bytes = c.rbuf.substr(0, total);
            c.rbuf.erase(0, total);
            staged_.push_back(std::move(req));
        }
        if (!staged_.empty()) {
            // one lock per readiness event, not per sliced request
            std::lock_guard<std::mutex> lk(mu_);
            // backstop: a protocol nobody harvests (e.g. HTTP/1.1 sent
            // to a gRPC-only listener) must not grow the queue without
            // bound — drop the connection instead
            if (ready_.size() + staged_.size() > (1u << 18)) {
                staged_.clear();
                return false;
            }
            for (auto& r : staged_) ready_.push_back(std::move(r));
            staged_.clear();
        }
        return true;
    }

    static void append_frame(std::string* w, uint8_t t, uint8_t f,
                             uint32_t sid, const char* p, size_t n) {
        char hdr[9];
        hdr[0] = (char)((n >> 16) & 0xFF);
        hdr[1] = (char)((n >> 8) & 0xFF);
        hdr[2] = (char)(n & 0xFF);
        hdr[3] = (char)t;
        hdr[4] = (char)f;
        hdr[5] = (char)((sid >> 24) & 0x7F);
        hdr[6] = (char)((sid >> 16) & 0xFF);
        hdr[7] = (char)((sid >> 8) & 0xFF);
        hdr[8] = (char)(sid & 0xFF);
        w->append(hdr, 9);
        if (n) w->append(p, n);
    }

    bool process_h2(Conn& c) {
        std::string& b = c.rbuf;
        size_t pos = 0;
        std::string resp;  // control frames to queue
        while (b.size() - pos >= 9) {
            const uint32_t len = ((uint8_t)b[pos] << 16) |
                                 ((uint8_t)b[pos + 1] << 8) |
                                 (uint8_t)b[pos + 2];
            if (len > (1u << 20)) return false;  // oversized frame
            if (b.size() - pos < 9 + (size_t)len) break;
            const uint8_t ftype = (uint8_t)b[pos + 3];
            const uint8_t flags = (uint8_t)b[pos + 4];
            const uint32_t sid = (((uint8_t)b[pos + 5] & 0x7F) << 24) |
                                 ((uint8_t)b[pos + 6] << 16) |
                                 ((uint8_t)b[pos + 7] << 8) |
                                 (uint8_t)b[pos + 8];
            const uint8_t* pl = (const uint8_t*)b.data() + pos + 9;
            size_t plen = len;
            H2State& h2 = *c.h2;
            switch (ftype) {
            case 0x4:  // SETTINGS
                if (!(flags & 0x1))
                    append_frame(&resp, 0x4, 0x1, 0, nullptr, 0);
                break;
            case 0x6:  // PING
                if (!(flags & 0x1))
                    append_frame(&resp, 0x6, 0x1, 0, (const char*)pl,
                                 plen);
                break;
            case 0x1: {  // HEADERS
                size_t off = 0;
                size_t pad = 0;
                if (flags & 0x8) pad = pl[off++];       // PADDED
                if (flags & 0x20) off += 5;             // PRIORITY
                if (off + pad > plen) return false;
                H2Stream& st = h2.streams[sid];
                st.hdr_frag.assign((const char*)pl + off,
                                   plen - off - pad);
                st.end_stream_pending = (flags & 0x1) != 0;
                if (flags & 0x4) {  // END_HEADERS
                    if (!finish_headers(c, sid,
                                        st.end_stream_pending))
                        return false;
                } else {
                    h2.cont_sid = sid;  // CONTINUATION completes it
                }
                break;
            }
            case 0x9: {  // CONTINUATION
                auto it = h2.streams.find(sid);
                if (it == h2.streams.end() || h2.cont_sid != sid)
                    return false;
                it->second.hdr_frag.append((const char*)pl, plen);
                if (flags & 0x4) {
                    h2.cont_sid = 0;
                    if (!finish_headers(c, sid,
                                        it->second.end_stream_pending))
                        return false;
                }
                break;
            }
            case 0x0: {  // DATA
                auto it = h2.streams.find(sid);
                if (it == h2.streams.end()) break;
                size_t off = 0, pad = 0;
                if (flags & 0x8) pad = pl[off++];
                if (off + pad > plen) return false;
                it->second.data.append((const char*)pl + off,
                                       plen - off - pad);
                if (plen) {  // replenish the connection recv window
                    const uint8_t wu[4] = {
                        (uint8_t)((plen >> 24) & 0x7F),
                        (uint8_t)(plen >> 16), (uint8_t)(plen >> 8),
                        (uint8_t)plen};
                    append_frame(&resp, 0x8, 0, 0, (const char*)wu, 4);
                }
                if (flags & 0x1) finish_stream(c, sid);
                break;
            }
            case 0x3:  // RST_STREAM
                h2.streams.erase(sid);
                break;
            case 0x7:  // GOAWAY
                c.close_after_write = true;
                break;
            default:
                break;  // WINDOW_UPDATE / PRIORITY / unknown: ignore
            }
            pos += 9 + len;
        }
        b.erase(0, pos);
        if (!resp.empty()) {
            std::lock_guard<std::recursive_mutex> lk(wmu_);
            c.wbuf += resp;
            pending_writes_.push_back(c.id);
        }
        return true;
    }

    bool finish_headers(Conn& c, uint32_t sid, bool end_stream) {
        H2State& h2 = *c.h2;
        H2Stream& st = h2.streams[sid];
        std::vector<std::pair<std::string, std::string>> hdrs;
        if (!h2.dec.decode((const uint8_t*)st.hdr_frag.data(),
                           st.hdr_frag.size(), &hdrs))
            return false;  // HPACK state is connection-fatal
        st.hdr_frag.clear();
        st.headers_done = true;
        st.path_id = -1;
        for (auto& kv : hdrs) {
            if (kv.first == ":path") {
                if (path_ids_) {
                    auto it = path_ids_->find(kv.second);
                    if (it != path_ids_->end()) st.path_id = it->second;
                }
                break;
            }
        }
        if (end_stream) finish_stream(c, sid);
        return true;
    }

    void finish_stream(Conn& c, uint32_t sid) {
        H2State& h2 = *c.h2;
        auto it = h2.streams.find(sid);
        if (it == h2.streams.end()) return;
        H2Stream& st = it->second;
        PendingGrpc g;
        g.conn_id = c.id;
        g.stream_id = sid;
        g.path_id = st.path_id;
        // strip the gRPC frame prefix: [compressed u8][len u32be]
        const std::string& d = st.data;
        if (d.size() >= 5 && d[0] == 0) {
            const uint32_t mlen = ((uint8_t)d[1] << 24) |
                                  ((uint8_t)d[2] << 16) |
                                  ((uint8_t)d[3] << 8) | (uint8_t)d[4];
            if (5 + (size_t)mlen <= d.size())
                g.msg = d.substr(5, mlen);
            else
                g.path_id = -2;
        } else {
            g.path_id = -2;  // compressed or malformed
        }
        h2.streams.erase(it);
        std::lock_guard<std::mutex> lk(mu_);
        if (gready_.size() > (1u << 18)) {
            c.close_after_write = true;  // unconsumed-protocol backstop
            return;
        }
        gready_.push_back(std::move(g));
    }

public:
    bool pop_grpc(PendingGrpc* out) {
        std::lock_guard<std::mutex> lk(mu_);
        if (gready_.empty()) return false;
        *out = std::move(gready_.front());
        gready_.pop_front();
        return true;
    }

    int pop_grpc_batch(std::vector<PendingGrpc>* out, int quota) {
        std::lock_guard<std::mutex> lk(mu_);
        int k = 0;
        while (k < quota && !gready_.empty()) {
            out->push_back(std::move(gready_.front()));
            gready_.pop_front();
            ++k;
        }
        return k;
    }

    // queue gRPC responses: HEADERS + DATA + trailers per row (rlen<0:
    // error -> headers + error trailers, no DATA)
    void queue_grpc_writes(const uint64_t* cids, const uint32_t* sids,
                           const int* rows, int nrows,
                           const uint8_t* out, const int32_t* roffs,
                           const int32_t* rlens,
                           const std::string& hdr_blk,
                           const std::string& trl_ok,
                           const std::string& trl_err) {
        std::lock_guard<std::recursive_mutex> lk(wmu_);
        for (int k = 0; k < nrows; ++k) {
            const int i = rows[k];
            auto it = conn_index_.find(cids[i]);
            if (it == conn_index_.end()) continue;
            std::string& w = it->second->wbuf;
            const uint32_t sid = sids[i];
            append_frame(&w, 0x1, 0x4, sid, hdr_blk.data(),
                         hdr_blk.size());
            if (rlens[i] >= 0) {
                append_frame(&w, 0x0, 0, sid,
                             (const char*)out + roffs[i],
                             (size_t)rlens[i]);
                append_frame(&w, 0x1, 0x4 | 0x1, sid, trl_ok.data(),
                             trl_ok.size());
            } else {
                append_frame(&w, 0x1, 0x4 | 0x1, sid, trl_err.data(),
                             trl_err.size());
            }
            pending_writes_.push_back(cids[i]);
        }
    }

    void set_path_ids(const std::unordered_map<std::string, int>* m) {
        path_ids_ = m;
    }

private:
    void drain_pending_writes() {
        std::vector<uint64_t> ids;
        {
            std::lock_guard<std::recursive_mutex> lk(wmu_);
            ids.swap(pending_writes_);
        }
        for (uint64_t id : ids) {
            auto it = conns_.find(id);
            if (it != conns_.end()) flush_conn(*it->second);
        }
    }

    void flush_conn(Conn& c) {
        std::lock_guard<std::recursive_mutex> lk(wmu_);
        while (!c.wbuf.empty()) {
            const ssize_t w = ::send(c.fd, c.wbuf.data(), c.wbuf.size(),
                                     MSG_NOSIGNAL);
            if (w < 0) {
                if (errno == EAGAIN || errno == EWOULDBLOCK) {
                    epoll_event ev{};
                    ev.events = EPOLLIN | EPOLLOUT;
                    ev.data.u64 = c.id;
                    epoll_ctl(ep_, EPOLL_CTL_MOD, c.fd, &ev);
                    return;
                }
                close_conn(c);
                return;
            }
            c.wbuf.erase(0, (size_t)w);
        }
        epoll_event ev{};
        ev.events = EPOLLIN;
        ev.data.u64 = c.id;
        epoll_ctl(ep_, EPOLL_CTL_MOD, c.fd, &ev);
        if (c.close_after_write) close_conn(c);
    }

    void close_conn(Conn& c) {
        if (c.dead) return;
        c.dead = true;
        epoll_ctl(ep_, EPOLL_CTL_DEL, c.fd, nullptr);
        ::close(c.fd);
        std::lock_guard<std::recursive_mutex> lk(wmu_);
        conn_index_.erase(c.id);
    }

    void gc_dead() {
        for (auto it = conns_.begin(); it != conns_.end();) {
            if (it->second->dead) it = conns_.erase(it);
            else ++it;
        }
    }

    int idx_;
    int port_;
    size_t max_req_;
    int listen_fd_ = -1;
    int ep_ = -1;
    std::atomic<bool> running_{false};
    std::thread loop_;
    uint64_t next_id_ = 1;
    // unique_ptr: Conn addresses must survive map rehash (conn_index_
    // and in-flight harvests hold raw pointers/ids)
    std::unordered_map<uint64_t, std::unique_ptr<Conn>> conns_;
    std::unordered_map<uint64_t, Conn*> conn_index_;
    std::mutex mu_;    // guards ready_
    // recursive: flush_conn (called under wmu_) may close_conn, which
    // locks wmu_ again. Guards conn_index_, pending_writes_ AND every
    // wbuf access (queue_writes appends from the engine thread while
    // the reactor flushes).
    std::recursive_mutex wmu_;
    std::deque<PendingReq> ready_;
    std::deque<PendingGrpc> gready_;  // completed unary gRPC requests
    std::vector<PendingReq> staged_;  // event-loop local slice buffer
    std::vector<uint64_t> pending_writes_;
    const std::unordered_map<std::string, int>* path_ids_ = nullptr;
};

// Persistent worker pool for the harvest/send fan-out: the serving
// thread's serial per-request work (ready-queue drain + egress routing)
// measured 150 + 120 ns/req — a ~3.6M req/s ceiling with the GPU at
// ~3 us/batch. run() executes fn(0..ntasks) across the workers + the
// calling thread and returns when all tasks finished.
class WorkerPool {
public:
    void start(int w) {
        nworkers_ = w < 1 ? 0 : w;
        running_ = true;
        for (int i = 0; i < nworkers_; ++i)
            threads_.emplace_back([this] { worker(); });
    }

    void stop() {
        {
            std::lock_guard<std::mutex> lk(mu_);
            if (!running_ && threads_.empty()) return;  // idempotent
            running_ = false;
            cv_.notify_all();
        }
        // drain: execute any queued tasks on this thread so a
        // concurrent run() caller can never wait forever on work the
        // exiting workers left behind
        while (true) {
            Task t;
            {
                std::lock_guard<std::mutex> lk(mu_);
                if (q_.empty()) break;
                t = q_.front();
                q_.pop_front();
            }
            exec(t);
        }
        for (auto& t : threads_) t.join();
        threads_.clear();
    }

    // Reentrant: several serving threads may run() concurrently (the
    // two-serve-thread loop overlaps one lane set's harvest with
    // another's egress); tasks queue globally and any participant may
    // execute any job's tasks (work conservation, no deadlock).
    void run(const std::function<void(int)>& fn, int ntasks) {
        if (nworkers_ == 0 || ntasks <= 1) {
            for (int i = 0; i < ntasks; ++i) fn(i);
            return;
        }
        Job job{&fn, ntasks};
        {
            std::lock_guard<std::mutex> lk(mu_);
            for (int i = 0; i < ntasks; ++i) q_.push_back({&job, i});
            cv_.notify_all();
        }
        // the caller works too (on any queued task)
        while (true) {
            Task t;
            {
                std::lock_guard<std::mutex> lk(mu_);
                if (q_.empty()) break;
                t = q_.front();
                q_.pop_front();
            }
            exec(t);
        }
        std::unique_lock<std::mutex> lk(mu_);
        cv_done_.wait(lk, [&] {
            return job.done.load(std::memory_order_acquire) >=
                   job.ntasks;
        });
    }

private:
    struct Job {
        const std::function<void(int)>* fn;
        int ntasks;
        std::atomic<int> done{0};
    };
    struct Task {
        Job* job = nullptr;
        int idx = 0;
    };

    void exec(const Task& t) {
        (*t.job->fn)(t.idx);
        if (t.job->done.fetch_add(1, std::memory_order_acq_rel) + 1 >=
            t.job->ntasks) {
            std::lock_guard<std::mutex> lk(mu_);
            cv_done_.notify_all();
        }
    }

    void worker() {
        while (true) {
            Task t;
            {
                std::unique_lock<std::mutex> lk(mu_);
                cv_.wait(lk, [&] { return !running_ || !q_.empty(); });
                if (!running_ && q_.empty()) return;
                t = q_.front();
                q_.pop_front();
            }
            exec(t);
        }
    }

    int nworkers_ = 0;
    std::vector<std::thread> threads_;
    std::mutex mu_;
    std::condition_variable cv_, cv_done_;
    std::deque<Task> q_;
    bool running_ = false;
};

class EpollServer {
public:
    EpollServer(int port, size_t max_req = 1 << 20, int threads = 4)
        : port_(port), max_req_(max_req),
          nthreads_(threads < 1 ? 1 : threads) {}

    ~EpollServer() { stop(); }

    // gRPC fast-path registration (BEFORE start): :path -> id
    int register_grpc_path(const std::string& path) {
        const int id = (int)grpc_paths_.size();
        grpc_paths_.push_back(path);
        path_ids_[path] = id;
        return id;
    }

    // precomputed HPACK blocks for the response frames (from the
    // Python HpackEncoder at server setup)
    void set_grpc_blocks(py::bytes hdr, py::bytes trl_ok,
                         py::bytes trl_err) {
        grpc_hdr_blk_ = std::string(hdr);
        grpc_trl_ok_ = std::string(trl_ok);
        grpc_trl_err_ = std::string(trl_err);
    }

    void start() {
        for (int i = 0; i < nthreads_; ++i) {
            reactors_.emplace_back(
                std::make_unique<Reactor>(i, port_, max_req_));
            port_ = reactors_.back()->bind_and_listen();  // 0 -> learned
            reactors_.back()->set_path_ids(&path_ids_);
        }
        for (auto& r : reactors_) r->run();
        // 3 workers + the calling thread measured best (more contend
        // on the reactor ready-queues); see profiles/SUMMARY.md r2
        const char* e = getenv("GOFR_IO_WORKERS");
        int w = e ? atoi(e) : 3;
        if (w > nthreads_ - 1) w = nthreads_ - 1;
        pool_.start(w);
    }

    void stop() {
        pool_.stop();
        for (auto& r : reactors_) r->stop();
        reactors_.clear();
    }

    int port() const { return port_; }

    // Harvest up to max_n complete requests into the caller's buffers
    // (the engine's pinned ingress ring). PARALLEL: reactor r's ready
    // queue drains into byte segment r of the ring (quota max_n/R
    // requests; leftovers stay queued for the next cycle) across the
    // worker pool, then one cheap serial pass merges the metadata
    // dense. The armed serving loop ships the whole fixed-size ring
    // every batch, so segmented placement costs no extra bus bytes.
    // Blocks up to window_us for the FIRST request. GIL released.
    // min_fill > 1 turns the window into a FILL deadline: keep
    // draining until min_fill requests arrived or window_us elapsed
    // (the adaptive batcher's throughput mode); min_fill <= 1 is the
    // latency-first return-on-first-drain behavior.
    std::pair<int, long> harvest(uintptr_t buf_ptr, long buf_cap,
                                 uintptr_t off_ptr, uintptr_t len_ptr,
                                 uintptr_t conn_ptr, int max_n,
                                 int window_us, int min_fill = 1) {
        py::gil_scoped_release rel;
        uint8_t* buf = (uint8_t*)buf_ptr;
        int64_t* offs = (int64_t*)off_ptr;
        int32_t* lens = (int32_t*)len_ptr;
        uint64_t* cids = (uint64_t*)conn_ptr;
        const int R = (int)reactors_.size();
        const long seg_bytes = buf_cap / R;
        const auto deadline = std::chrono::steady_clock::now() +
                              std::chrono::microseconds(window_us);
        int n = 0;
        long nbytes = 0;
        if (max_n < R) {
            // tiny batches: serial drain (quota math needs max_n >= R)
            while (true) {
                for (auto& r : reactors_) {
                    PendingReq req;
                    while (n < max_n &&
                           r->pop_ready_fit(&req, buf_cap - nbytes)) {
                        memcpy(buf + nbytes, req.bytes.data(),
                               req.bytes.size());
                        offs[n] = nbytes;
                        lens[n] = (int32_t)req.bytes.size();
                        cids[n] = req.conn_id;
                        nbytes += (long)req.bytes.size();
                        ++n;
                    }
                }
                if (n > 0 ||
                    std::chrono::steady_clock::now() >= deadline)
                    return {n, nbytes};
                std::this_thread::sleep_for(
                    std::chrono::microseconds(50));
            }
        }
        const int quota = max_n / R;
        // per-serving-thread scratch (two serve threads harvest
        // concurrently into their own lanes)
        static thread_local std::vector<Scratch> scratch_tl;
        std::vector<Scratch>& scratch_ = scratch_tl;
        scratch_.resize(R);
        while (true) {
            pool_.run([&](int r) {
                Scratch& s = scratch_[r];
                s.offs.resize(quota);
                s.lens.resize(quota);
                s.cids.resize(quota);
                s.reqs.clear();
                reactors_[r]->pop_ready_batch(&s.reqs, quota,
                                              seg_bytes);
                s.n = 0;
                const long base = (long)r * seg_bytes;
                long pos = 0;
                for (PendingReq& req : s.reqs) {
                    memcpy(buf + base + pos, req.bytes.data(),
                           req.bytes.size());
                    s.offs[s.n] = base + pos;
                    s.lens[s.n] = (int32_t)req.bytes.size();
                    s.cids[s.n] = req.conn_id;
                    pos += (long)req.bytes.size();
                    ++s.n;
                }
                s.used = pos;
            }, R);
            for (int r = 0; r < R; ++r) {
                Scratch& s = scratch_[r];
                if (!s.n) continue;
                memcpy(offs + n, s.offs.data(), sizeof(int64_t) * s.n);
                memcpy(lens + n, s.lens.data(), sizeof(int32_t) * s.n);
                memcpy(cids + n, s.cids.data(), sizeof(uint64_t) * s.n);
                n += s.n;
                const long end = (long)r * seg_bytes + s.used;
                if (end > nbytes) nbytes = end;
            }
            if (n >= (min_fill > 1 ? min_fill : 1) ||
                (n > 0 && min_fill <= 1) ||
                std::chrono::steady_clock::now() >= deadline)
                break;
            std::this_thread::sleep_for(std::chrono::microseconds(50));
        }
        return {n, nbytes};
    }

    // Write responses back (engine egress pinned buffer), routed to the
    // owning reactor by the conn-id's reactor index; the per-reactor
    // egress work (index lookup, close-scan, wbuf append) fans out
    // across the worker pool.
    void send(uintptr_t conn_ptr, int n, uintptr_t out_ptr,
              uintptr_t roff_ptr, uintptr_t rlen_ptr) {
        py::gil_scoped_release rel;
        const uint64_t* cids = (const uint64_t*)conn_ptr;
        const uint8_t* out = (const uint8_t*)out_ptr;
        const int32_t* roffs = (const int32_t*)roff_ptr;
        const int32_t* rlens = (const int32_t*)rlen_ptr;
        const int R = (int)reactors_.size();
        std::vector<std::vector<int>> by_reactor(R);
        for (int i = 0; i < n; ++i) {
            const int r = id_reactor(cids[i]);
            if (r >= 0 && r < R && rlens[i] > 0)
                by_reactor[r].push_back(i);
        }
        pool_.run([&](int r) {
            if (!by_reactor[r].empty())
                reactors_[r]->queue_writes(cids, by_reactor[r].data(),
                                           (int)by_reactor[r].size(),
                                           out, roffs, rlens);
        }, R);
    }

    long ready_count() {
        long t = 0;
        for (auto& r : reactors_) t += r->ready_count();
        return t;
    }

    // Harvest completed unary gRPC request messages (packed bytes +
    // per-message conn id / stream id / registered path id). Blocks up
    // to window_us for the first message. GIL released.
    std::pair<int, long> harvest_grpc(uintptr_t buf_ptr, long buf_cap,
                                      uintptr_t off_ptr,
                                      uintptr_t len_ptr,
                                      uintptr_t cid_ptr,
                                      uintptr_t sid_ptr,
                                      uintptr_t pid_ptr, int max_n,
                                      int window_us) {
        py::gil_scoped_release rel;
        uint8_t* buf = (uint8_t*)buf_ptr;
        int64_t* offs = (int64_t*)off_ptr;
        int32_t* lens = (int32_t*)len_ptr;
        uint64_t* cids = (uint64_t*)cid_ptr;
        uint32_t* sids = (uint32_t*)sid_ptr;
        int32_t* pids = (int32_t*)pid_ptr;
        const auto deadline = std::chrono::steady_clock::now() +
                              std::chrono::microseconds(window_us);
        int n = 0;
        long pos = 0;
        std::vector<PendingGrpc> batch;
        while (true) {
            for (auto& r : reactors_) {
                batch.clear();
                r->pop_grpc_batch(&batch, max_n - n);
                for (PendingGrpc& g : batch) {
                    if (pos + (long)g.msg.size() > buf_cap) break;
                    memcpy(buf + pos, g.msg.data(), g.msg.size());
                    offs[n] = pos;
                    lens[n] = (int32_t)g.msg.size();
                    cids[n] = g.conn_id;
                    sids[n] = g.stream_id;
                    pids[n] = g.path_id;
                    pos += (long)g.msg.size();
                    ++n;
                }
                if (n >= max_n) break;
            }
            if (n > 0 || std::chrono::steady_clock::now() >= deadline)
                break;
            std::this_thread::sleep_for(std::chrono::microseconds(50));
        }
        return {n, pos};
    }

    // Write gRPC responses: per row HEADERS + DATA (the complete gRPC
    // response frame bytes) + trailers, routed by conn id. rlen < 0
    // renders the error-trailers variant.
    void send_grpc(uintptr_t cid_ptr, uintptr_t sid_ptr, int n,
                   uintptr_t out_ptr, uintptr_t roff_ptr,
                   uintptr_t rlen_ptr) {
        py::gil_scoped_release rel;
        const uint64_t* cids = (const uint64_t*)cid_ptr;
        const uint32_t* sids = (const uint32_t*)sid_ptr;
        const uint8_t* out = (const uint8_t*)out_ptr;
        const int32_t* roffs = (const int32_t*)roff_ptr;
        const int32_t* rlens = (const int32_t*)rlen_ptr;
        const int R = (int)reactors_.size();
        std::vector<std::vector<int>> by_reactor(R);
        for (int i = 0; i < n; ++i) {
            const int r = id_reactor(cids[i]);
            if (r >= 0 && r < R) by_reactor[r].push_back(i);
        }
        pool_.run([&](int r) {
            if (!by_reactor[r].empty())
                reactors_[r]->queue_grpc_writes(
                    cids, sids, by_reactor[r].data(),
                    (int)by_reactor[r].size(), out, roffs, rlens,
                    grpc_hdr_blk_, grpc_trl_ok_, grpc_trl_err_);
        }, R);
    }

    // Sharded harvest (multi-GPU serving): drain ready requests into a
    // SLOT-layout exchange buffer of `world` blocks x `bpr` slots x
    // `slot` bytes, the fixed-size all-to-all granularity of
    // engine.shard.AllToAllSharder. A request's block = its OWNER rank
    // = splitmix-hash(conn_id) % world (stable conn->owner affinity).
    // Overfull blocks spill to a backlog drained first next cycle.
    // lens[] is zeroed for unfilled slots (FL_EMPTY padding on device);
    // conn ids for unfilled slots are 0 (never allocated). Requests
    // larger than `slot` are dropped (the sharded exchange is
    // fixed-granularity; size `slot` for the workload).
    int harvest_slots(uintptr_t buf_ptr, int slot, int world, int bpr,
                      uintptr_t len_ptr, uintptr_t conn_ptr,
                      int window_us) {
        py::gil_scoped_release rel;
        uint8_t* buf = (uint8_t*)buf_ptr;
        int32_t* lens = (int32_t*)len_ptr;
        uint64_t* cids = (uint64_t*)conn_ptr;
        const int total = world * bpr;
        memset(lens, 0, sizeof(int32_t) * (size_t)total);
        memset(cids, 0, sizeof(uint64_t) * (size_t)total);
        std::vector<int> fill(world, 0);
        int n = 0;
        auto place = [&](PendingReq& req) -> bool {
            if ((long)req.bytes.size() > slot) return true;  // dropped
            const uint64_t h = req.conn_id * 0x9E3779B97F4A7C15ull;
            const int owner = (int)((h >> 32) % (uint64_t)world);
            if (fill[owner] >= bpr) {
                backlog_.push_back(std::move(req));
                return false;  // block full: spilled
            }
            const int s = owner * bpr + fill[owner]++;
            memcpy(buf + (size_t)s * slot, req.bytes.data(),
                   req.bytes.size());
            lens[s] = (int32_t)req.bytes.size();
            cids[s] = req.conn_id;
            ++n;
            return true;
        };
        const auto deadline = std::chrono::steady_clock::now() +
                              std::chrono::microseconds(window_us);
        {
            // backlog first (already popped from the reactors)
            std::deque<PendingReq> keep;
            while (!backlog_.empty()) {
                PendingReq req = std::move(backlog_.front());
                backlog_.pop_front();
                const uint64_t h = req.conn_id * 0x9E3779B97F4A7C15ull;
                const int owner = (int)((h >> 32) % (uint64_t)world);
                if (fill[owner] >= bpr || (long)req.bytes.size() > slot) {
                    if ((long)req.bytes.size() <= slot)
                        keep.push_back(std::move(req));
                    continue;
                }
                place(req);
            }
            backlog_.swap(keep);
        }
        while (true) {
            for (auto& r : reactors_) {
                while (true) {
                    PendingReq req;
                    if (!r->pop_ready(&req)) break;
                    place(req);
                }
            }
            if (n > 0 || std::chrono::steady_clock::now() >= deadline)
                break;
            std::this_thread::sleep_for(std::chrono::microseconds(50));
        }
        return n;
    }

private:
    struct Scratch {
        std::vector<int64_t> offs;
        std::vector<int32_t> lens;
        std::vector<uint64_t> cids;
        std::vector<PendingReq> reqs;
        int n = 0;
        long used = 0;
    };

    int port_;
    size_t max_req_;
    int nthreads_;
    std::deque<PendingReq> backlog_;
    std::vector<std::unique_ptr<Reactor>> reactors_;
    WorkerPool pool_;
    std::unordered_map<std::string, int> path_ids_;
    std::vector<std::string> grpc_paths_;
    std::string grpc_hdr_blk_, grpc_trl_ok_, grpc_trl_err_;
};

}  // namespace

#ifndef GOFR_SRC_HASH
#define GOFR_SRC_HASH "unhashed"
#endif

// RESP2 bulk-array parse (the redis MGET reply): item spans into the
// caller's buffer, no copies. Returns (n_items_parsed, complete) —
// offsets/lens arrays must hold max_items int64/int32; a nil bulk
// ($-1) records len == -1. Python-side reply parsing measured ~1 us
// per item (readline loops); this is the batched-datasource path's
// host bottleneck remover.
static py::tuple resp_parse_array(uintptr_t buf_ptr, long nbytes,
                                  long max_items, uintptr_t off_ptr,
                                  uintptr_t len_ptr) {
    const char* b = (const char*)buf_ptr;
    int64_t* offs = (int64_t*)off_ptr;
    int32_t* lens = (int32_t*)len_ptr;
    long pos = 0;
    auto line_end = [&](long p) -> long {
        const void* e = memchr(b + p, '\n', (size_t)(nbytes - p));
        return e ? (long)((const char*)e - b) : -1;
    };
    if (nbytes < 4 || b[0] != '*') return py::make_tuple(0, false);
    long le = line_end(0);
    if (le < 0) return py::make_tuple(0, false);
    long count = strtol(b + 1, nullptr, 10);
    if (count < 0) count = 0;
    if (count > max_items) throw std::runtime_error("resp: too many");
    pos = le + 1;
    long n = 0;
    while (n < count) {
        if (pos >= nbytes || b[pos] != '$')
            return py::make_tuple(0, false);
        le = line_end(pos);
        if (le < 0) return py::make_tuple(0, false);
        const long blen = strtol(b + pos + 1, nullptr, 10);
        pos = le + 1;
        if (blen < 0) {
            offs[n] = pos;
            lens[n] = -1;  // nil bulk
        } else {
            if (pos + blen + 2 > nbytes) return py::make_tuple(0, false);
            offs[n] = pos;
            lens[n] = (int32_t)blen;
            pos += blen + 2;
        }
        ++n;
    }
    return py::make_tuple((long)n, true);
}

PYBIND11_MODULE(_core, m) {
    m.def("resp_parse_array", &resp_parse_array);
    m.doc() = "gofr_amd native epoll ingress (multi-reactor)";
    // source-hash stamp (tests/test_build_hash.py): the loaded binary
    // must carry the hash of the source committed in the tree
    m.attr("src_hash") = GOFR_SRC_HASH;
    py::class_<EpollServer>(m, "EpollServer")
        .def(py::init<int, size_t, int>(), py::arg("port"),
             py::arg("max_req") = (size_t)1 << 20,
             py::arg("threads") = 4)
        .def("start", &EpollServer::start)
        .def("stop", &EpollServer::stop)
        .def("port", &EpollServer::port)
        .def("ready_count", &EpollServer::ready_count)
        .def("harvest", &EpollServer::harvest,
             py::arg("buf_ptr"), py::arg("buf_cap"), py::arg("off_ptr"),
             py::arg("len_ptr"), py::arg("conn_ptr"), py::arg("max_n"),
             py::arg("window_us"), py::arg("min_fill") = 1)
        .def("harvest_slots", &EpollServer::harvest_slots)
        .def("send", &EpollServer::send)
        .def("register_grpc_path", &EpollServer::register_grpc_path)
        .def("set_grpc_blocks", &EpollServer::set_grpc_blocks)
        .def("harvest_grpc", &EpollServer::harvest_grpc)
        .def("send_grpc", &EpollServer::send_grpc);
}
