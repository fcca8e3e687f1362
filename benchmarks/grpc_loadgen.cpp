// Pipelined h2c gRPC loopback load generator (config 3's client side,
// socket-attached). Speaks the minimal client half of HTTP/2: preface,
// SETTINGS, a wide WINDOW_UPDATE, then `depth` outstanding unary
// SayHello streams per connection (HEADERS + DATA END_STREAM with
// raw-literal HPACK, constant bytes per request except the stream id).
// Completion = trailers HEADERS with END_STREAM. Reports msgs/s + p50/
// p99 latency as one JSON line.
//
// usage: grpc_loadgen <host> <port> <conns> <depth> <seconds> <threads>

#include <arpa/inet.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

static double now_s() {
    using namespace std::chrono;
    return duration<double>(steady_clock::now().time_since_epoch())
        .count();
}

static void frame_hdr(std::string* w, uint8_t t, uint8_t f, uint32_t sid,
                      size_t n) {
    char h[9];
    h[0] = (char)((n >> 16) & 0xFF);
    h[1] = (char)((n >> 8) & 0xFF);
    h[2] = (char)(n & 0xFF);
    h[3] = (char)t;
    h[4] = (char)f;
    h[5] = (char)((sid >> 24) & 0x7F);
    h[6] = (char)((sid >> 16) & 0xFF);
    h[7] = (char)((sid >> 8) & 0xFF);
    h[8] = (char)(sid & 0xFF);
    w->append(h, 9);
}

// literal-without-indexing HPACK (lengths < 127 assumed)
static void lit(std::string* b, const char* n, const char* v) {
    b->push_back(0x00);
    b->push_back((char)strlen(n));
    b->append(n);
    b->push_back((char)strlen(v));
    b->append(v);
}

struct Conn {
    int fd = -1;
    std::string rbuf;
    uint32_t next_sid = 1;
    int outstanding = 0;
    std::vector<double> sent_at;  // ring indexed by (sid>>1) % depth
};

struct Stats {
    long done = 0;
    std::vector<float> lat_us;
};

static void reactor(const char* host, int port, int nconns, int depth,
                    double seconds, Stats* st) {
    // precompute the per-request byte blob (sid patched per send)
    std::string hdr_blk;
    lit(&hdr_blk, ":method", "POST");
    lit(&hdr_blk, ":scheme", "http");
    lit(&hdr_blk, ":path", "/hello.HelloService/SayHello");
    lit(&hdr_blk, ":authority", "localhost");
    lit(&hdr_blk, "content-type", "application/grpc");
    lit(&hdr_blk, "te", "trailers");
    // HelloRequest{name:"bench-client"}: field 1 len-delim
    const char name[] = "bench-client";
    std::string msg;
    msg.push_back(0x0A);
    msg.push_back((char)strlen(name));
    msg.append(name);
    std::string grpc_frame;
    grpc_frame.push_back(0);
    const uint32_t ml = (uint32_t)msg.size();
    grpc_frame.push_back((char)(ml >> 24));
    grpc_frame.push_back((char)(ml >> 16));
    grpc_frame.push_back((char)(ml >> 8));
    grpc_frame.push_back((char)ml);
    grpc_frame += msg;
    std::string unit;  // HEADERS + DATA, sid at fixed offsets
    frame_hdr(&unit, 0x1, 0x4, 0, hdr_blk.size());
    unit += hdr_blk;
    const size_t data_off = unit.size();
    frame_hdr(&unit, 0x0, 0x1, 0, grpc_frame.size());  // END_STREAM
    unit += grpc_frame;

    int ep = epoll_create1(0);
    std::vector<Conn> conns(nconns);
    std::string pre = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";
    frame_hdr(&pre, 0x4, 0, 0, 0);  // SETTINGS
    const uint8_t wu[4] = {0x3f, 0xff, 0xff, 0xff};
    frame_hdr(&pre, 0x8, 0, 0, 4);
    pre.append((const char*)wu, 4);
    for (int i = 0; i < nconns; ++i) {
        int fd = socket(AF_INET, SOCK_STREAM, 0);
        sockaddr_in a{};
        a.sin_family = AF_INET;
        a.sin_port = htons(port);
        inet_pton(AF_INET, host, &a.sin_addr);
        if (connect(fd, (sockaddr*)&a, sizeof(a)) != 0) {
            perror("connect");
            exit(2);
        }
        int one = 1;
        setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
        send(fd, pre.data(), pre.size(), MSG_NOSIGNAL);
        // non-blocking AFTER connect+preface: the read loop polls
        const int fl = fcntl(fd, F_GETFL, 0);
        fcntl(fd, F_SETFL, fl | O_NONBLOCK);
        conns[i].fd = fd;
        conns[i].sent_at.resize((size_t)depth * 2);
        epoll_event ev{};
        ev.events = EPOLLIN;
        ev.data.u32 = (uint32_t)i;
        epoll_ctl(ep, EPOLL_CTL_ADD, fd, &ev);
    }
    auto pump = [&](Conn& c) {  // fill to depth outstanding
        std::string burst;
        while (c.outstanding < depth) {
            const uint32_t sid = c.next_sid;
            c.next_sid += 2;
            std::string u = unit;
            u[5] = (char)((sid >> 24) & 0x7F);
            u[6] = (char)((sid >> 16) & 0xFF);
            u[7] = (char)((sid >> 8) & 0xFF);
            u[8] = (char)(sid & 0xFF);
            u[data_off + 5] = (char)((sid >> 24) & 0x7F);
            u[data_off + 6] = (char)((sid >> 16) & 0xFF);
            u[data_off + 7] = (char)((sid >> 8) & 0xFF);
            u[data_off + 8] = (char)(sid & 0xFF);
            c.sent_at[(sid >> 1) % c.sent_at.size()] = now_s();
            burst += u;
            ++c.outstanding;
        }
        if (!burst.empty())
            send(c.fd, burst.data(), burst.size(), MSG_NOSIGNAL);
    };
    for (auto& c : conns) pump(c);
    const double t_end = now_s() + seconds;
    std::vector<epoll_event> events(256);
    std::string tmp(1 << 16, '\0');
    while (now_s() < t_end) {
        const int k = epoll_wait(ep, events.data(), (int)events.size(),
                                 50);
        for (int e = 0; e < k; ++e) {
            Conn& c = conns[events[e].data.u32];
            while (true) {
                const ssize_t r = recv(c.fd, tmp.data(), tmp.size(), 0);
                if (r <= 0) break;
                c.rbuf.append(tmp.data(), (size_t)r);
            }
            size_t pos = 0;
            while (c.rbuf.size() - pos >= 9) {
                const uint32_t len =
                    ((uint8_t)c.rbuf[pos] << 16) |
                    ((uint8_t)c.rbuf[pos + 1] << 8) |
                    (uint8_t)c.rbuf[pos + 2];
                if (c.rbuf.size() - pos < 9 + (size_t)len) break;
                const uint8_t ft = (uint8_t)c.rbuf[pos + 3];
                const uint8_t fl = (uint8_t)c.rbuf[pos + 4];
                const uint32_t sid =
                    (((uint8_t)c.rbuf[pos + 5] & 0x7F) << 24) |
                    ((uint8_t)c.rbuf[pos + 6] << 16) |
                    ((uint8_t)c.rbuf[pos + 7] << 8) |
                    (uint8_t)c.rbuf[pos + 8];
                if (ft == 0x1 && (fl & 0x1) && sid) {  // trailers
                    ++st->done;
                    --c.outstanding;
                    const double dt =
                        now_s() -
                        c.sent_at[(sid >> 1) % c.sent_at.size()];
                    if (st->lat_us.size() < (1u << 20))
                        st->lat_us.push_back((float)(dt * 1e6));
                }
                pos += 9 + len;
            }
            c.rbuf.erase(0, pos);
            pump(c);
        }
    }
    for (auto& c : conns) close(c.fd);
    close(ep);
}

int main(int argc, char** argv) {
    if (argc < 7) {
        fprintf(stderr, "usage: %s host port conns depth secs threads\n",
                argv[0]);
        return 1;
    }
    const char* host = argv[1];
    const int port = atoi(argv[2]);
    const int conns = atoi(argv[3]);
    const int depth = atoi(argv[4]);
    const double secs = atof(argv[5]);
    const int threads = atoi(argv[6]);
    std::vector<Stats> st(threads);
    std::vector<std::thread> ts;
    const double t0 = now_s();
    for (int t = 0; t < threads; ++t)
        ts.emplace_back(reactor, host, port,
                        std::max(1, conns / threads), depth, secs,
                        &st[t]);
    for (auto& t : ts) t.join();
    const double el = now_s() - t0;
    long done = 0;
    std::vector<float> lat;
    for (auto& s : st) {
        done += s.done;
        lat.insert(lat.end(), s.lat_us.begin(), s.lat_us.end());
    }
    std::sort(lat.begin(), lat.end());
    const double p50 = lat.empty() ? 0 : lat[lat.size() / 2];
    const double p99 = lat.empty() ? 0 : lat[(size_t)(lat.size() * 0.99)];
    printf("{\"msg_per_s\": %.1f, \"done\": %ld, \"seconds\": %.2f, "
           "\"conns\": %d, \"depth\": %d, \"threads\": %d, "
           "\"p50_us\": %.1f, \"p99_us\": %.1f}\n",
           done / el, done, el, conns, depth, threads, p50, p99);
    return 0;
}
