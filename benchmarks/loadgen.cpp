// Pipelined HTTP loopback load generator (config 1's client side).
// The Python client caps around ~0.5M req/s on the GIL; this epoll
// client keeps `depth` requests outstanding per connection across
// `threads` reactor threads and reports req/s + latency percentiles.
//
// usage: loadgen <host> <port> <conns> <depth> <seconds> <threads>
// output: one JSON line on stdout.

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

static const char REQ[] =
    "GET /greet HTTP/1.1\r\nHost: localhost\r\n\r\n";

struct Conn {
    int fd = -1;
    std::string rbuf;
    size_t need_body = 0;   // body bytes still to consume
    bool in_body = false;
    std::vector<double> sent_at;  // ring of send timestamps
    size_t sent_head = 0, sent_tail = 0;
    size_t outstanding = 0;
};

static double now_s() {
    using namespace std::chrono;
    return duration<double>(steady_clock::now().time_since_epoch()).count();
}

struct Stats {
    long done = 0;
    std::vector<float> lat_us;
};

static void reactor(const char* host, int port, int nconns, int depth,
                    double seconds, Stats* st, int n_addrs) {
    int ep = epoll_create1(0);
    std::vector<Conn> conns(nconns);
    for (int i = 0; i < nconns; ++i) {
        int fd = socket(AF_INET, SOCK_STREAM, 0);
        sockaddr_in a{};
        a.sin_family = AF_INET;
        a.sin_port = htons(port);
        if (n_addrs > 1) {
            // spread over loopback addresses so >64k conns fit inside
            // the ephemeral-port space (config 5: 100k real conns)
            char ab[32];
            snprintf(ab, sizeof(ab), "127.0.0.%d", 1 + (i % n_addrs));
            inet_pton(AF_INET, ab, &a.sin_addr);
        } else {
            inet_pton(AF_INET, host, &a.sin_addr);
        }
        if (connect(fd, (sockaddr*)&a, sizeof(a))) { perror("connect"); _exit(2); }
        int one = 1;
        setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
        conns[i].fd = fd;
        conns[i].sent_at.resize(depth + 1);
        epoll_event ev{};
        ev.events = EPOLLIN;
        ev.data.u32 = i;
        epoll_ctl(ep, EPOLL_CTL_ADD, fd, &ev);
    }
    const size_t rlen = sizeof(REQ) - 1;
    // prime: fill every connection's pipeline
    std::string burst;
    for (int d = 0; d < depth; ++d) burst += REQ;
    const double t0 = now_s();
    for (auto& c : conns) {
        (void)!write(c.fd, burst.data(), burst.size());
        for (int d = 0; d < depth; ++d)
            c.sent_at[(c.sent_tail++) % c.sent_at.size()] = t0;
        c.outstanding = depth;
    }
    std::vector<char> buf(1 << 16);
    epoll_event evs[64];
    const double deadline = t0 + seconds;
    st->lat_us.reserve(1 << 20);
    while (now_s() < deadline) {
        int nev = epoll_wait(ep, evs, 64, 50);
        for (int e = 0; e < nev; ++e) {
            Conn& c = conns[evs[e].data.u32];
            ssize_t got = read(c.fd, buf.data(), buf.size());
            if (got <= 0) { fprintf(stderr, "conn closed\n"); _exit(3); }
            c.rbuf.append(buf.data(), got);
            int completed = 0;
            for (;;) {
                if (c.in_body) {
                    if (c.rbuf.size() < c.need_body) break;
                    c.rbuf.erase(0, c.need_body);
                    c.in_body = false;
                    ++completed;
                    continue;
                }
                size_t he = c.rbuf.find("\r\n\r\n");
                if (he == std::string::npos) break;
                size_t clp = c.rbuf.find("Content-Length:");
                size_t cl = 0;
                if (clp != std::string::npos && clp < he)
                    cl = strtoul(c.rbuf.c_str() + clp + 15, nullptr, 10);
                c.rbuf.erase(0, he + 4);
                c.need_body = cl;
                c.in_body = true;
            }
            if (completed) {
                const double now = now_s();
                for (int k = 0; k < completed; ++k) {
                    const double ts =
                        c.sent_at[(c.sent_head++) % c.sent_at.size()];
                    if (st->lat_us.size() < st->lat_us.capacity())
                        st->lat_us.push_back(float((now - ts) * 1e6));
                }
                st->done += completed;
                c.outstanding -= completed;
                // refill the pipeline
                std::string out;
                for (int k = 0; k < completed; ++k) out += REQ;
                (void)!write(c.fd, out.data(), out.size());
                for (int k = 0; k < completed; ++k)
                    c.sent_at[(c.sent_tail++) % c.sent_at.size()] = now;
                c.outstanding += completed;
                (void)rlen;
            }
        }
    }
    for (auto& c : conns) close(c.fd);
    close(ep);
}

int main(int argc, char** argv) {
    if (argc < 7) {
        fprintf(stderr,
                "usage: %s host port conns depth seconds threads "
                "[n_addrs]\n", argv[0]);
        return 1;
    }
    const char* host = argv[1];
    int port = atoi(argv[2]), conns = atoi(argv[3]), depth = atoi(argv[4]);
    double seconds = atof(argv[5]);
    int threads = atoi(argv[6]);
    int n_addrs = argc > 7 ? atoi(argv[7]) : 1;
    std::vector<Stats> st(threads);
    std::vector<std::thread> ts;
    const double t0 = now_s();
    int per = conns / threads;
    for (int t = 0; t < threads; ++t)
        ts.emplace_back(reactor, host, port, per, depth, seconds, &st[t],
                        n_addrs);
    for (auto& t : ts) t.join();
    const double elapsed = now_s() - t0;
    long total = 0;
    std::vector<float> all;
    for (auto& s : st) {
        total += s.done;
        all.insert(all.end(), s.lat_us.begin(), s.lat_us.end());
    }
    std::sort(all.begin(), all.end());
    auto pct = [&](double p) {
        return all.empty() ? 0.f : all[size_t(p * (all.size() - 1))];
    };
    printf("{\"req_per_s\": %.1f, \"total\": %ld, \"seconds\": %.2f, "
           "\"p50_us\": %.1f, \"p99_us\": %.1f, \"conns\": %d, "
           "\"depth\": %d, \"threads\": %d}\n",
           total / elapsed, total, elapsed, pct(0.50), pct(0.99),
           conns, depth, threads);
    return 0;
}
