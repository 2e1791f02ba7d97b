// forge_hey — native closed-loop HTTP load generator (hey analog).
//
// Reference analog: tests/hey/hey.sh (10k requests / 200 concurrency,
// POST /rpc) and the 1000-user locust knee (crates/mcp_runtime/STATUS.md).
// The round-1 rig used aiohttp client processes, which saturate near
// ~8-10k RPS each and measure themselves; a C++ epoll client keeps the
// generator off the critical path when the server serves 10^5+ RPS.
//
// Step protocol (bench.py drives it):
//   * connect C connections across T threads
//   * run W warmup steps of R requests each, print "WARM" and wait for a
//     "GO" line on stdin (bench.py runs its barrier + cuda sync in between)
//   * run K timed steps of R requests each; per-step wall time measured
//     here; per-request latencies recorded during timed steps only
//   * print one JSON line with totals + latency percentiles, exit 0
//
// Each connection is closed-loop: send request, read full response, send
// the next. A step ends when R requests have completed; connections idle
// at a step boundary until the next step starts (the payload sequence is
// rotated so varying-argument traffic exercises the pipeline, matching
// bench.py's generator).

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <time.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <fstream>
#include <memory>
#include <mutex>
#include <sstream>
#include <string>
#include <thread>
#include <vector>

namespace {

double mono_s() {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return (double)ts.tv_sec + ts.tv_nsec * 1e-9;
}

struct Config {
    std::string host = "127.0.0.1";
    int port = 8080;
    std::string path = "/rpc";
    int connections = 200;
    int threads = 4;
    long requests_per_step = 10000;
    int warmup_steps = 1;
    int steps = 5;
    std::vector<std::string> payloads;  // rotated round-robin
    std::string auth;                   // Authorization header value
    bool handshake = true;              // WARM/GO coordination
};

struct Conn {
    int fd = -1;
    std::string out;      // request bytes not yet written
    size_t out_off = 0;
    std::string in;       // response accumulation
    size_t need = 0;      // body bytes still required (after header)
    size_t header_len = 0;
    bool have_header = false;
    double t_send = 0.0;
    bool busy = false;
};

struct Shared {
    Config* cfg;
    std::atomic<long> remaining{0};   // requests left to LAUNCH this step
    std::atomic<long> inflight{0};
    std::atomic<long> done{0};        // responses completed this step
    std::atomic<long> errors{0};
    std::atomic<long> non200{0};
    std::atomic<int> phase{0};        // step counter; -1 = exit
    std::atomic<bool> record{false};  // record latencies (timed steps)
    std::mutex lat_mu;
    std::vector<uint32_t> lat_us;
    // step barrier
    std::mutex mu;
    std::condition_variable cv;
    long step_target = 0;
};

int connect_to(const Config& c) {
    int fd = socket(AF_INET, SOCK_STREAM, IPPROTO_TCP);
    if (fd < 0) return -1;
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    struct sockaddr_in addr;
    memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)c.port);
    inet_pton(AF_INET, c.host.c_str(), &addr.sin_addr);
    if (connect(fd, (struct sockaddr*)&addr, sizeof(addr)) != 0) {
        close(fd);
        return -1;
    }
    fcntl(fd, F_SETFL, fcntl(fd, F_GETFL, 0) | O_NONBLOCK);
    return fd;
}

std::string build_request(const Config& c, const std::string& payload) {
    std::ostringstream o;
    o << "POST " << c.path << " HTTP/1.1\r\nHost: " << c.host << "\r\n"
      << "Content-Type: application/json\r\nContent-Length: " << payload.size() << "\r\n";
    if (!c.auth.empty()) o << "Authorization: " << c.auth << "\r\n";
    o << "Connection: keep-alive\r\n\r\n" << payload;
    return o.str();
}

struct Worker {
    Shared* sh;
    const Config* cfg;
    int tid;
    int epfd = -1;
    std::vector<Conn> conns;
    std::vector<std::string> reqs;  // prebuilt request bytes, rotated
    size_t req_idx = 0;
    std::vector<uint32_t> local_lat;

    void start_one(Conn& c, int idx) {
        long rem = sh->remaining.fetch_sub(1);
        if (rem <= 0) {
            sh->remaining.fetch_add(1);
            return;
        }
        c.out = reqs[req_idx];
        req_idx = (req_idx + 1) % reqs.size();
        c.out_off = 0;
        c.in.clear();
        c.have_header = false;
        c.t_send = mono_s();
        c.busy = true;
        sh->inflight++;
        pump_write(c, idx);
    }

    void rearm(Conn& c, int idx, uint32_t events) {
        struct epoll_event ev;
        ev.events = events;
        ev.data.u32 = (uint32_t)idx;
        epoll_ctl(epfd, EPOLL_CTL_MOD, c.fd, &ev);
    }

    void reconnect(Conn& c, int idx) {
        epoll_ctl(epfd, EPOLL_CTL_DEL, c.fd, nullptr);
        close(c.fd);
        c.fd = connect_to(*cfg);
        if (c.fd < 0) {
            sh->errors++;
            return;
        }
        struct epoll_event ev;
        ev.events = EPOLLIN;
        ev.data.u32 = (uint32_t)idx;
        epoll_ctl(epfd, EPOLL_CTL_ADD, c.fd, &ev);
    }

    void fail_request(Conn& c, int idx) {
        sh->errors++;
        if (c.busy) {
            c.busy = false;
            sh->inflight--;
            sh->done++;
        }
        reconnect(c, idx);
    }

    void pump_write(Conn& c, int idx) {
        while (c.out_off < c.out.size()) {
            ssize_t w = write(c.fd, c.out.data() + c.out_off, c.out.size() - c.out_off);
            if (w > 0) {
                c.out_off += (size_t)w;
                continue;
            }
            if (w < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
                rearm(c, idx, EPOLLIN | EPOLLOUT);
                return;
            }
            if (w < 0 && errno == EINTR) continue;
            fail_request(c, idx);
            return;
        }
        rearm(c, idx, EPOLLIN);
    }

    void pump_read(Conn& c, int idx) {
        char buf[65536];
        for (;;) {
            ssize_t r = read(c.fd, buf, sizeof(buf));
            if (r > 0) {
                c.in.append(buf, (size_t)r);
                if (try_complete(c, idx)) return;
                if (r < (ssize_t)sizeof(buf)) return;
                continue;
            }
            if (r == 0) {  // server closed
                fail_request(c, idx);
                return;
            }
            if (errno == EAGAIN || errno == EWOULDBLOCK) return;
            if (errno == EINTR) continue;
            fail_request(c, idx);
            return;
        }
    }

    // returns true if a response completed (and possibly a new request began)
    bool try_complete(Conn& c, int idx) {
        if (!c.busy) {
            c.in.clear();
            return false;
        }
        if (!c.have_header) {
            const char* p = (const char*)memmem(c.in.data(), c.in.size(), "\r\n\r\n", 4);
            if (p == nullptr) return false;
            c.header_len = (size_t)(p - c.in.data()) + 4;
            // status
            int status = 0;
            if (c.in.size() > 12) status = atoi(c.in.data() + 9);
            if (status != 200 && status != 202) sh->non200++;
            // content-length
            size_t cl = 0;
            const char* h = c.in.data();
            const char* hend = h + c.header_len;
            const char* q = h;
            while (q < hend) {
                const char* le = (const char*)memchr(q, '\r', (size_t)(hend - q));
                if (!le) break;
                if (le - q > 15 && strncasecmp(q, "content-length:", 15) == 0) {
                    cl = (size_t)strtoull(q + 15, nullptr, 10);
                    break;
                }
                q = le + 2;
            }
            c.need = cl;
            c.have_header = true;
        }
        if (c.in.size() < c.header_len + c.need) return false;
        // response complete
        double lat = mono_s() - c.t_send;
        if (sh->record.load(std::memory_order_relaxed))
            local_lat.push_back((uint32_t)(lat * 1e6));
        c.busy = false;
        sh->inflight--;
        c.in.erase(0, c.header_len + c.need);
        c.have_header = false;
        long d = ++sh->done;
        if (d >= sh->step_target) {
            sh->cv.notify_all();  // step complete
        } else {
            start_one(c, idx);
        }
        return true;
    }

    void run() {
        epfd = epoll_create1(0);
        int per = cfg->connections / cfg->threads + (tid < cfg->connections % cfg->threads ? 1 : 0);
        conns.resize((size_t)per);
        for (int i = 0; i < per; ++i) {
            conns[(size_t)i].fd = connect_to(*cfg);
            if (conns[(size_t)i].fd < 0) {
                sh->errors++;
                continue;
            }
            struct epoll_event ev;
            ev.events = EPOLLIN;
            ev.data.u32 = (uint32_t)i;
            epoll_ctl(epfd, EPOLL_CTL_ADD, conns[(size_t)i].fd, &ev);
        }
        // rotate payloads with a per-thread offset so threads don't send
        // identical sequences
        for (size_t i = 0; i < cfg->payloads.size(); ++i)
            reqs.push_back(build_request(*cfg, cfg->payloads[i]));
        req_idx = (size_t)tid % reqs.size();

        int last_phase = 0;
        struct epoll_event evs[128];
        for (;;) {
            int phase = sh->phase.load(std::memory_order_acquire);
            if (phase < 0) break;
            if (phase != last_phase) {
                last_phase = phase;
                // kick every idle connection
                for (int i = 0; i < per; ++i) {
                    Conn& c = conns[(size_t)i];
                    if (c.fd >= 0 && !c.busy) start_one(c, i);
                    if (sh->remaining.load(std::memory_order_relaxed) <= 0) break;
                }
            }
            int n = epoll_wait(epfd, evs, 128, 5);
            for (int i = 0; i < n; ++i) {
                int idx = (int)evs[i].data.u32;
                Conn& c = conns[(size_t)idx];
                if (c.fd < 0) continue;
                if (evs[i].events & (EPOLLHUP | EPOLLERR)) {
                    fail_request(c, idx);
                    continue;
                }
                if (evs[i].events & EPOLLOUT) pump_write(c, idx);
                if (evs[i].events & EPOLLIN) pump_read(c, idx);
            }
        }
        if (!local_lat.empty()) {
            std::lock_guard<std::mutex> g(sh->lat_mu);
            sh->lat_us.insert(sh->lat_us.end(), local_lat.begin(), local_lat.end());
        }
        for (auto& c : conns)
            if (c.fd >= 0) close(c.fd);
        close(epfd);
    }
};

double pct(std::vector<uint32_t>& v, double p) {
    if (v.empty()) return 0.0;
    size_t k = (size_t)(p * (double)(v.size() - 1));
    std::nth_element(v.begin(), v.begin() + (long)k, v.end());
    return (double)v[k] / 1000.0;  // ms
}

}  // namespace

int main(int argc, char** argv) {
    Config cfg;
    std::string payload_file;
    for (int i = 1; i < argc; ++i) {
        std::string a = argv[i];
        auto next = [&]() -> const char* { return i + 1 < argc ? argv[++i] : ""; };
        if (a == "--host") cfg.host = next();
        else if (a == "--port") cfg.port = atoi(next());
        else if (a == "--path") cfg.path = next();
        else if (a == "--connections" || a == "-c") cfg.connections = atoi(next());
        else if (a == "--threads" || a == "-t") cfg.threads = atoi(next());
        else if (a == "--requests-per-step" || a == "-r") cfg.requests_per_step = atol(next());
        else if (a == "--warmup" || a == "-w") cfg.warmup_steps = atoi(next());
        else if (a == "--steps" || a == "-k") cfg.steps = atoi(next());
        else if (a == "--payload") cfg.payloads.push_back(next());
        else if (a == "--payload-file") payload_file = next();
        else if (a == "--auth") cfg.auth = next();
        else if (a == "--no-handshake") cfg.handshake = false;
        else {
            fprintf(stderr, "unknown arg %s\n", a.c_str());
            return 2;
        }
    }
    if (!payload_file.empty()) {  // one JSON payload per line
        std::ifstream f(payload_file);
        std::string line;
        while (std::getline(f, line))
            if (!line.empty()) cfg.payloads.push_back(line);
    }
    if (cfg.payloads.empty())
        cfg.payloads.push_back(
            "{\"jsonrpc\":\"2.0\",\"id\":1,\"method\":\"ping\"}");
    if (cfg.threads < 1) cfg.threads = 1;
    if (cfg.connections < cfg.threads) cfg.connections = cfg.threads;

    Shared sh;
    sh.cfg = &cfg;
    std::vector<std::unique_ptr<Worker>> workers;
    std::vector<std::thread> threads;
    for (int t = 0; t < cfg.threads; ++t) {
        auto w = std::make_unique<Worker>();
        w->sh = &sh;
        w->cfg = &cfg;
        w->tid = t;
        workers.push_back(std::move(w));
    }
    for (auto& w : workers) threads.emplace_back([&w] { w->run(); });

    auto run_step = [&](bool record) -> double {
        sh.done = 0;
        sh.step_target = cfg.requests_per_step;
        sh.remaining = cfg.requests_per_step;
        sh.record = record;
        double t0 = mono_s();
        sh.phase.fetch_add(1, std::memory_order_release);
        std::unique_lock<std::mutex> g(sh.mu);
        // poll-wait: the notify is fired without sh.mu held, so re-check
        // on a short cadence rather than trusting a single wakeup
        while (sh.done.load() < sh.step_target && mono_s() - t0 < 600.0)
            sh.cv.wait_for(g, std::chrono::milliseconds(10));
        return mono_s() - t0;
    };

    for (int s = 0; s < cfg.warmup_steps; ++s) run_step(false);
    if (cfg.handshake) {
        printf("WARM\n");
        fflush(stdout);
        char line[64];
        if (fgets(line, sizeof(line), stdin) == nullptr) return 3;  // expect GO
    }
    std::vector<double> step_s;
    double t0 = mono_s();
    for (int s = 0; s < cfg.steps; ++s) step_s.push_back(run_step(true));
    double elapsed = mono_s() - t0;
    sh.phase = -1;
    for (auto& th : threads) th.join();

    long total = (long)cfg.steps * cfg.requests_per_step;
    double p50 = pct(sh.lat_us, 0.50), p90 = pct(sh.lat_us, 0.90), p99 = pct(sh.lat_us, 0.99);
    printf("{\"requests\":%ld,\"elapsed_s\":%.6f,\"rps\":%.2f,"
           "\"p50_ms\":%.3f,\"p90_ms\":%.3f,\"p99_ms\":%.3f,"
           "\"errors\":%ld,\"non200\":%ld,\"connections\":%d,\"threads\":%d,"
           "\"steps\":%d,\"requests_per_step\":%ld,\"max_step_s\":%.6f}\n",
           total, elapsed, (double)total / elapsed, p50, p90, p99,
           sh.errors.load(), sh.non200.load(), cfg.connections, cfg.threads,
           cfg.steps, cfg.requests_per_step, *std::max_element(step_s.begin(), step_s.end()));
    fflush(stdout);
    return 0;
}
