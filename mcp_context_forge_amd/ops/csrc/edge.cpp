// Native HTTP edge: the GPU-owner's socket loop in C++.
//
// Reference analog: crates/mcp_runtime (the Rust axum sidecar that owns the
// public /mcp ingress, lib.rs:1347 build_router) — here the native edge IS
// the gateway process's front door, not a sidecar: N epoll threads accept,
// parse HTTP/1.1, authenticate against an in-process credential cache, and
// stage whole request batches for Python (the engine + GPU pipeline).
// Python sees batches, never sockets.
//
// Round-1 measurement that motivates this: the asyncio owner loop saturated
// at ~31-35k RPS with 6 Python worker shells feeding it (profiles/README.md
// worker-count diagnostic) — per-request Python/asyncio work WAS the
// ceiling. Here the per-request host work (parse, auth memo, response
// framing, socket IO) is native and threaded; Python's cost is per-BATCH.
//
// Threading model
//   * T epoll threads, each with its own SO_REUSEPORT listener (the kernel
//     load-balances connections), level-triggered epoll, non-blocking fds.
//   * One global request queue feeds Python: edge_poll() blocks (GIL
//     released) on a condition variable and drains up to max_n requests.
//   * Completions (edge_complete) run on the Python caller's thread: format
//     the HTTP response, append to the connection's output buffer under its
//     mutex, flush; EAGAIN / close / resume-parse work is bounced to the
//     owning epoll thread over an eventfd.
//   * HTTP/1.1 keep-alive; one request in flight per connection (a
//     pipelining client is buffered, not corrupted: parsing pauses while a
//     request is in flight and resumes on completion).
//
// Auth: authorization-header -> interned-user cache. Hits answer from C++
// (no Python); misses ride the batch with the raw header so Python
// authenticates once and edge_auth_put()s the result. Negative entries
// expire after NEG_TTL so a not-yet-minted token is not refused forever;
// edge_auth_clear() drops everything on revocation.

#ifndef _GNU_SOURCE
#define _GNU_SOURCE
#endif
#define PY_SSIZE_T_CLEAN
#include <Python.h>

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <stdint.h>
#include <string.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <time.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <deque>
#include <memory>
#include <mutex>
#include <shared_mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

namespace {

constexpr size_t MAX_HEADER = 16 * 1024;
constexpr double NEG_TTL = 2.0;  // seconds a failed-auth memo is trusted
constexpr uint64_t EV_WAKE = UINT64_MAX;
constexpr uint64_t EV_LISTEN = UINT64_MAX - 1;

double mono_s() {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return (double)ts.tv_sec + ts.tv_nsec * 1e-9;
}

enum ReqKind : uint8_t { K_RPC = 0, K_OTHER = 1 };

struct Conn {
    std::mutex mu;
    int fd = -1;
    uint32_t slot = 0;
    uint32_t gen = 1;          // bumped on close; completions check it
    bool closed = true;
    bool want_close = false;   // close after flush
    bool epollout = false;     // EPOLLOUT currently armed
    int inflight = 0;          // requests handed to Python, not yet answered
    std::string in;            // unconsumed input bytes
    std::string out;           // unflushed output bytes
    // current-request parse state
    bool have_header = false;
    size_t header_len = 0;
    size_t body_need = 0;
    bool keep_alive = true;
    uint8_t kind = K_RPC;
    std::string method, target, authz, headers_blob;
};

struct Req {
    uint64_t id = 0;
    uint8_t kind = K_RPC;
    std::string body;
    const std::string* user = nullptr;  // interned; null = unresolved
    std::string authz;                  // raw header value when unresolved
    std::string method, target, headers_blob;  // K_OTHER only
};

struct EdgeThread;

struct Edge {
    int port = 0;
    int nthreads = 0;
    size_t max_body = 4u << 20;
    bool auth_required = true;
    std::atomic<bool> stop{false};
    std::vector<std::unique_ptr<EdgeThread>> threads;

    std::mutex qmu;
    std::condition_variable qcv;
    std::deque<Req> queue;

    std::shared_mutex auth_mu;
    struct AuthEntry {
        const std::string* user = nullptr;  // null = negative entry
        double neg_until = 0.0;
    };
    std::unordered_map<std::string, AuthEntry> auth;
    std::deque<std::unique_ptr<std::string>> interned;

    std::atomic<uint64_t> accepted{0}, closed_conns{0}, hot{0}, cold{0},
        direct_401{0}, direct_health{0}, responses{0}, bytes_in{0}, bytes_out{0},
        parse_errors{0};
};

struct EdgeThread {
    Edge* e = nullptr;
    int tid = 0;
    int epfd = -1;
    int evfd = -1;
    int listen_fd = -1;
    std::vector<Conn*> conns;          // slot-indexed; never freed (gen reuse)
    std::vector<uint32_t> free_slots;
    std::mutex todo_mu;
    std::vector<uint32_t> todo;        // slots needing epoll-thread attention
    std::thread th;

    ~EdgeThread() {
        for (Conn* c : conns) delete c;
        if (epfd >= 0) ::close(epfd);
        if (evfd >= 0) ::close(evfd);
        if (listen_fd >= 0) ::close(listen_fd);
    }

    void wake_with(uint32_t slot) {
        {
            std::lock_guard<std::mutex> g(todo_mu);
            todo.push_back(slot);
        }
        uint64_t one = 1;
        ssize_t rc = ::write(evfd, &one, 8);
        (void)rc;
    }

    void wake_with_many(std::vector<uint32_t>&& slots) {
        if (slots.empty()) return;
        {
            std::lock_guard<std::mutex> g(todo_mu);
            todo.insert(todo.end(), slots.begin(), slots.end());
        }
        uint64_t one = 1;
        ssize_t rc = ::write(evfd, &one, 8);
        (void)rc;
    }
};

uint64_t make_id(int tid, uint32_t slot, uint32_t gen) {
    return ((uint64_t)(unsigned)tid << 56) | ((uint64_t)(gen & 0xFFFFFF) << 32) | slot;
}

const char* reason_of(int code) {
    switch (code) {
        case 200: return "OK";
        case 202: return "Accepted";
        case 400: return "Bad Request";
        case 401: return "Unauthorized";
        case 403: return "Forbidden";
        case 404: return "Not Found";
        case 413: return "Payload Too Large";
        case 429: return "Too Many Requests";
        case 431: return "Request Header Fields Too Large";
        case 501: return "Not Implemented";
        default: return "Internal Server Error";
    }
}

void format_response(std::string& out, int code, const char* body, size_t n, bool keep_alive) {
    char head[160];
    int hn = snprintf(head, sizeof(head),
                      "HTTP/1.1 %d %s\r\nContent-Type: application/json\r\n"
                      "Content-Length: %zu\r\nConnection: %s\r\n\r\n",
                      code, reason_of(code), n, keep_alive ? "keep-alive" : "close");
    out.append(head, (size_t)hn);
    out.append(body, n);
}

// flush c->out on the socket; arm EPOLLOUT on partial. caller holds c->mu.
// Safe from any thread: epoll_ctl is thread-safe, and the epollout flag is
// guarded by c->mu.
void flush_locked(Edge* e, EdgeThread* t, Conn* c) {
    while (!c->out.empty()) {
        ssize_t w = ::write(c->fd, c->out.data(), c->out.size());
        if (w > 0) {
            e->bytes_out += (uint64_t)w;
            c->out.erase(0, (size_t)w);
            continue;
        }
        if (w < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
            if (!c->epollout) {
                c->epollout = true;
                struct epoll_event ev;
                ev.events = EPOLLIN | EPOLLOUT;
                ev.data.u64 = c->slot;
                epoll_ctl(t->epfd, EPOLL_CTL_MOD, c->fd, &ev);
            }
            return;
        }
        if (w < 0 && errno == EINTR) continue;
        c->want_close = true;  // write error — drop buffered output
        c->out.clear();
        return;
    }
    if (c->epollout) {
        c->epollout = false;
        struct epoll_event ev;
        ev.events = EPOLLIN;
        ev.data.u64 = c->slot;
        epoll_ctl(t->epfd, EPOLL_CTL_MOD, c->fd, &ev);
    }
}

struct EdgeLoop {
    Edge* e;
    EdgeThread* t;

    void close_slot(uint32_t slot) {  // epoll thread only
        Conn* c = t->conns[slot];
        {
            std::lock_guard<std::mutex> g(c->mu);
            if (c->closed) return;
            epoll_ctl(t->epfd, EPOLL_CTL_DEL, c->fd, nullptr);
            ::close(c->fd);
            c->closed = true;
            c->gen++;
            c->in.clear();
            c->out.clear();
            c->inflight = 0;
            c->have_header = false;
            c->epollout = false;
        }
        e->closed_conns++;
        t->free_slots.push_back(slot);
    }

    void direct_respond_locked(Conn* c, int code, const char* body) {
        bool ka = c->keep_alive && code < 400;
        if (code == 401) ka = c->keep_alive;  // 401 keeps the conn (clients retry)
        format_response(c->out, code, body, strlen(body), ka);
        if (!ka) c->want_close = true;
        flush_locked(e, t, c);
    }

    bool parse_header(Conn* c) {
        const char* base = c->in.data();
        const char* end = base + c->header_len - 2;  // before final CRLF
        const char* eol = (const char*)memchr(base, '\r', c->header_len);
        if (!eol) return false;
        const char* sp1 = (const char*)memchr(base, ' ', (size_t)(eol - base));
        if (!sp1) return false;
        const char* sp2 = (const char*)memchr(sp1 + 1, ' ', (size_t)(eol - sp1 - 1));
        if (!sp2) return false;
        c->method.assign(base, (size_t)(sp1 - base));
        c->target.assign(sp1 + 1, (size_t)(sp2 - sp1 - 1));
        c->keep_alive = true;  // HTTP/1.1 default
        c->body_need = 0;
        c->authz.clear();
        bool expect_continue = false, chunked = false;
        const char* line = eol + 2;
        while (line < end) {
            const char* le = (const char*)memchr(line, '\r', (size_t)(end - line));
            if (!le) le = end;
            const char* colon = (const char*)memchr(line, ':', (size_t)(le - line));
            if (colon) {
                size_t kn = (size_t)(colon - line);
                const char* v = colon + 1;
                while (v < le && *v == ' ') v++;
                size_t vn = (size_t)(le - v);
                if (kn == 14 && strncasecmp(line, "content-length", 14) == 0) {
                    c->body_need = (size_t)strtoull(std::string(v, vn).c_str(), nullptr, 10);
                } else if (kn == 13 && strncasecmp(line, "authorization", 13) == 0) {
                    c->authz.assign(v, vn);
                } else if (kn == 10 && strncasecmp(line, "connection", 10) == 0) {
                    if (vn == 5 && strncasecmp(v, "close", 5) == 0) c->keep_alive = false;
                } else if (kn == 6 && strncasecmp(line, "expect", 6) == 0) {
                    expect_continue = vn >= 3 && strncasecmp(v, "100", 3) == 0;
                } else if (kn == 17 && strncasecmp(line, "transfer-encoding", 17) == 0) {
                    chunked = true;
                }
            }
            line = le + 2;
        }
        bool is_rpc_path = c->target.size() >= 4 && memcmp(c->target.data(), "/rpc", 4) == 0 &&
                           (c->target.size() == 4 || c->target[4] == '?');
        bool is_health = c->method == "GET" &&
                         (c->target == "/healthz" || c->target == "/health");
        if ((c->method == "POST" && is_rpc_path) || is_health) {
            c->kind = K_RPC;
        } else {
            c->kind = K_OTHER;
            c->headers_blob.assign(base, c->header_len);
        }
        if (chunked) return false;  // the edge requires Content-Length
        if (expect_continue) c->out += "HTTP/1.1 100 Continue\r\n\r\n";
        return true;
    }

    // parse as many complete requests as allowed; caller holds c->mu
    void parse_and_dispatch(Conn* c) {
        std::vector<Req> ready;
        while (c->inflight == 0 && !c->closed && !c->want_close) {
            if (!c->have_header) {
                if (c->in.size() < 4) break;
                const char* p = (const char*)memmem(c->in.data(), c->in.size(), "\r\n\r\n", 4);
                if (p == nullptr) {
                    if (c->in.size() > MAX_HEADER) {
                        e->parse_errors++;
                        direct_respond_locked(c, 431, "{\"detail\":\"header too large\"}");
                    }
                    break;
                }
                c->header_len = (size_t)(p - c->in.data()) + 4;
                if (!parse_header(c)) {
                    e->parse_errors++;
                    direct_respond_locked(c, 400, "{\"detail\":\"malformed request\"}");
                    break;
                }
                if (c->body_need > e->max_body) {
                    direct_respond_locked(c, 413, "{\"detail\":\"request body too large\"}");
                    break;
                }
                c->have_header = true;
                if (!c->out.empty()) flush_locked(e, t, c);  // e.g. 100 Continue
            }
            if (c->in.size() < c->header_len + c->body_need) break;  // need more body

            std::string body = c->in.substr(c->header_len, c->body_need);
            c->in.erase(0, c->header_len + c->body_need);
            c->have_header = false;

            if (c->kind == K_RPC && c->method == "GET") {
                e->direct_health++;
                direct_respond_locked(c, 200, "{\"status\":\"ok\",\"edge\":\"native\"}");
                continue;
            }
            const std::string* user = nullptr;
            if (c->kind == K_RPC) {
                if (c->authz.empty()) {
                    if (e->auth_required) {
                        e->direct_401++;
                        direct_respond_locked(c, 401, "{\"detail\":\"Not authenticated\"}");
                        continue;
                    }
                } else {
                    bool deny = false;
                    {
                        std::shared_lock<std::shared_mutex> g(e->auth_mu);
                        auto it = e->auth.find(c->authz);
                        if (it != e->auth.end()) {
                            if (it->second.user != nullptr) user = it->second.user;
                            else if (it->second.neg_until > mono_s()) deny = true;
                        }
                    }
                    if (deny) {
                        e->direct_401++;
                        direct_respond_locked(c, 401, "{\"detail\":\"Not authenticated\"}");
                        continue;
                    }
                }
            }
            Req r;
            r.id = make_id(t->tid, c->slot, c->gen);
            r.kind = c->kind;
            r.body = std::move(body);
            r.user = user;
            if (c->kind == K_RPC) {
                if (user == nullptr) r.authz = c->authz;
                e->hot++;
            } else {
                r.method = c->method;
                r.target = c->target;
                r.headers_blob = std::move(c->headers_blob);
                e->cold++;
            }
            c->inflight++;
            ready.push_back(std::move(r));
        }
        if (!ready.empty()) {
            {
                std::lock_guard<std::mutex> g(e->qmu);
                for (auto& r : ready) e->queue.push_back(std::move(r));
            }
            e->qcv.notify_one();
        }
    }

    bool read_into(Conn* c) {
        char buf[65536];
        for (;;) {
            ssize_t r = ::read(c->fd, buf, sizeof(buf));
            if (r > 0) {
                e->bytes_in += (uint64_t)r;
                c->in.append(buf, (size_t)r);
                if (r < (ssize_t)sizeof(buf)) return true;
                continue;
            }
            if (r == 0) return false;  // peer closed
            if (errno == EAGAIN || errno == EWOULDBLOCK) return true;
            if (errno == EINTR) continue;
            return false;
        }
    }

    void accept_loop() {
        for (;;) {
            int fd = accept4(t->listen_fd, nullptr, nullptr, SOCK_NONBLOCK);
            if (fd < 0) return;
            int one = 1;
            setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
            uint32_t slot;
            if (!t->free_slots.empty()) {
                slot = t->free_slots.back();
                t->free_slots.pop_back();
            } else {
                slot = (uint32_t)t->conns.size();
                t->conns.push_back(new Conn());
                t->conns[slot]->slot = slot;
            }
            Conn* c = t->conns[slot];
            {
                std::lock_guard<std::mutex> g(c->mu);
                c->fd = fd;
                c->closed = false;
                c->want_close = false;
                c->epollout = false;
                c->inflight = 0;
                c->in.clear();
                c->out.clear();
                c->have_header = false;
                c->keep_alive = true;
            }
            struct epoll_event ev;
            ev.events = EPOLLIN;
            ev.data.u64 = slot;
            epoll_ctl(t->epfd, EPOLL_CTL_ADD, fd, &ev);
            e->accepted++;
        }
    }

    void handle_todo(uint32_t slot) {
        Conn* c = t->conns[slot];
        bool do_close = false;
        {
            std::lock_guard<std::mutex> g(c->mu);
            if (c->closed) return;
            flush_locked(e, t, c);
            if (!c->in.empty() && c->inflight == 0) parse_and_dispatch(c);
            if (c->want_close && c->out.empty() && c->inflight == 0) do_close = true;
        }
        if (do_close) close_slot(slot);
    }

    void run() {
        struct epoll_event evs[256];
        while (!e->stop.load(std::memory_order_relaxed)) {
            int n = epoll_wait(t->epfd, evs, 256, 100);
            for (int i = 0; i < n; ++i) {
                uint64_t d = evs[i].data.u64;
                if (d == EV_WAKE) {
                    uint64_t v;
                    while (read(t->evfd, &v, 8) == 8) {}
                    std::vector<uint32_t> todo;
                    {
                        std::lock_guard<std::mutex> g(t->todo_mu);
                        todo.swap(t->todo);
                    }
                    for (uint32_t slot : todo) handle_todo(slot);
                    continue;
                }
                if (d == EV_LISTEN) {
                    accept_loop();
                    continue;
                }
                uint32_t slot = (uint32_t)d;
                Conn* c = t->conns[slot];
                bool do_close = false;
                {
                    std::lock_guard<std::mutex> g(c->mu);
                    if (c->closed) continue;
                    if (evs[i].events & (EPOLLHUP | EPOLLERR)) {
                        do_close = c->inflight == 0;
                        if (!do_close) c->want_close = true;
                    } else {
                        if (evs[i].events & EPOLLOUT) flush_locked(e, t, c);
                        if (evs[i].events & EPOLLIN) {
                            if (!read_into(c)) {
                                // peer half-closed: finish in-flight work, no new reads
                                if (c->inflight == 0 && c->out.empty()) do_close = true;
                                else c->want_close = true;
                            } else {
                                parse_and_dispatch(c);
                            }
                        }
                        if (c->want_close && c->out.empty() && c->inflight == 0) do_close = true;
                    }
                }
                if (do_close) close_slot(slot);
            }
        }
        for (uint32_t s = 0; s < t->conns.size(); ++s) {
            Conn* c = t->conns[s];
            std::lock_guard<std::mutex> g(c->mu);
            if (!c->closed) {
                epoll_ctl(t->epfd, EPOLL_CTL_DEL, c->fd, nullptr);
                ::close(c->fd);
                c->closed = true;
                c->gen++;
            }
        }
    }
};

int make_listener(int port, int backlog) {
    int fd = socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK, IPPROTO_TCP);
    if (fd < 0) return -1;
    int one = 1;
    setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    setsockopt(fd, SOL_SOCKET, SO_REUSEPORT, &one, sizeof(one));
    struct sockaddr_in addr;
    memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_ANY);
    addr.sin_port = htons((uint16_t)port);
    if (bind(fd, (struct sockaddr*)&addr, sizeof(addr)) != 0 || listen(fd, backlog) != 0) {
        ::close(fd);
        return -1;
    }
    return fd;
}

Edge* edge_start_impl(int port, int nthreads, size_t max_body, bool auth_required, int backlog) {
    auto* e = new Edge();
    e->port = port;
    e->nthreads = nthreads;
    e->max_body = max_body;
    e->auth_required = auth_required;
    for (int i = 0; i < nthreads; ++i) {
        auto t = std::make_unique<EdgeThread>();
        t->e = e;
        t->tid = i;
        t->listen_fd = make_listener(port, backlog);
        if (t->listen_fd < 0) {
            e->stop = true;
            for (auto& tt : e->threads)
                if (tt->th.joinable()) tt->th.join();
            delete e;
            return nullptr;
        }
        t->epfd = epoll_create1(0);
        t->evfd = eventfd(0, EFD_NONBLOCK);
        struct epoll_event ev;
        ev.events = EPOLLIN;
        ev.data.u64 = EV_WAKE;
        epoll_ctl(t->epfd, EPOLL_CTL_ADD, t->evfd, &ev);
        ev.events = EPOLLIN;
        ev.data.u64 = EV_LISTEN;
        epoll_ctl(t->epfd, EPOLL_CTL_ADD, t->listen_fd, &ev);
        EdgeThread* tp = t.get();
        t->th = std::thread([e, tp]() {
            EdgeLoop loop{e, tp};
            loop.run();
        });
        e->threads.push_back(std::move(t));
    }
    return e;
}

// complete one response: returns false if the connection is gone.
// When `deferred_wake` is non-null the response is only appended to the
// connection's output buffer and the slot recorded per thread — the owning
// epoll threads then flush everything with ONE eventfd wake per thread,
// parallelizing the per-connection write() syscalls across the edge
// threads instead of serializing them on the completer thread (measured:
// the serial flush was ~10 ms of a 2.6k-row batch at saturation).
bool complete_one(Edge* e, uint64_t id, int status, const char* body, size_t n, bool raw,
                  std::vector<std::vector<uint32_t>>* deferred_wake = nullptr) {
    int tid = (int)(id >> 56);
    uint32_t gen = (uint32_t)((id >> 32) & 0xFFFFFF);
    uint32_t slot = (uint32_t)(id & 0xFFFFFFFF);
    if (tid < 0 || tid >= (int)e->threads.size()) return false;
    EdgeThread* t = e->threads[(size_t)tid].get();
    if (slot >= t->conns.size()) return false;
    Conn* c = t->conns[slot];
    bool need_wake = false;
    {
        std::lock_guard<std::mutex> g(c->mu);
        if (c->closed || (c->gen & 0xFFFFFF) != gen) return false;
        if (raw) c->out.append(body, n);
        else format_response(c->out, status, body, n, c->keep_alive);
        if (!c->keep_alive) c->want_close = true;
        c->inflight--;
        if (deferred_wake != nullptr) {
            (*deferred_wake)[(size_t)tid].push_back(slot);
        } else {
            flush_locked(e, t, c);
            // epoll thread must take over when: more buffered input to
            // parse, unflushed output pending, or the conn should close
            need_wake = (!c->in.empty() && c->inflight == 0) || c->epollout ||
                        (c->want_close && c->inflight == 0);
        }
    }
    e->responses++;
    if (need_wake) t->wake_with(slot);
    return true;
}

// ------------------------------------------------------------- Python module

PyObject* py_edge_start(PyObject*, PyObject* args) {
    int port, nthreads, auth_required, backlog = 4096;
    unsigned long long max_body;
    if (!PyArg_ParseTuple(args, "iiKp|i", &port, &nthreads, &max_body, &auth_required, &backlog))
        return nullptr;
    Edge* e = edge_start_impl(port, nthreads, (size_t)max_body, auth_required != 0, backlog);
    if (e == nullptr) {
        PyErr_Format(PyExc_OSError, "edge: cannot bind port %d", port);
        return nullptr;
    }
    return PyLong_FromVoidPtr(e);
}

PyObject* py_edge_poll(PyObject*, PyObject* args) {
    unsigned long long handle;
    int wait_us, linger_us, max_n;
    if (!PyArg_ParseTuple(args, "Kiii", &handle, &wait_us, &linger_us, &max_n)) return nullptr;
    Edge* e = (Edge*)(uintptr_t)handle;
    std::vector<Req> batch;
    Py_BEGIN_ALLOW_THREADS;
    {
        std::unique_lock<std::mutex> g(e->qmu);
        if (e->queue.empty() && wait_us > 0) {
            e->qcv.wait_for(g, std::chrono::microseconds(wait_us),
                            [&] { return !e->queue.empty() || e->stop.load(); });
        }
        if (!e->queue.empty() && linger_us > 0 && (int)e->queue.size() < max_n) {
            g.unlock();
            std::this_thread::sleep_for(std::chrono::microseconds(linger_us));
            g.lock();
        }
        int n = (int)e->queue.size();
        if (n > max_n) n = max_n;
        batch.reserve((size_t)n);
        for (int i = 0; i < n; ++i) {
            batch.push_back(std::move(e->queue.front()));
            e->queue.pop_front();
        }
    }
    Py_END_ALLOW_THREADS;

    size_t n = batch.size();
    PyObject* ids = PyBytes_FromStringAndSize(nullptr, (Py_ssize_t)(n * 8));
    PyObject* kinds = PyBytes_FromStringAndSize(nullptr, (Py_ssize_t)n);
    PyObject* bodies = PyList_New((Py_ssize_t)n);
    PyObject* users = PyList_New((Py_ssize_t)n);
    PyObject* authzs = PyList_New((Py_ssize_t)n);
    PyObject* meta = PyList_New((Py_ssize_t)n);
    if (!ids || !kinds || !bodies || !users || !authzs || !meta) return nullptr;
    uint64_t* idp = (uint64_t*)PyBytes_AS_STRING(ids);
    char* kp = PyBytes_AS_STRING(kinds);
    for (size_t i = 0; i < n; ++i) {
        Req& r = batch[i];
        idp[i] = r.id;
        kp[i] = (char)r.kind;
        PyList_SET_ITEM(bodies, (Py_ssize_t)i,
                        PyBytes_FromStringAndSize(r.body.data(), (Py_ssize_t)r.body.size()));
        if (r.user != nullptr) {
            PyList_SET_ITEM(users, (Py_ssize_t)i,
                            PyUnicode_FromStringAndSize(r.user->data(), (Py_ssize_t)r.user->size()));
        } else {
            Py_INCREF(Py_None);
            PyList_SET_ITEM(users, (Py_ssize_t)i, Py_None);
        }
        if (!r.authz.empty()) {
            PyList_SET_ITEM(authzs, (Py_ssize_t)i,
                            PyBytes_FromStringAndSize(r.authz.data(), (Py_ssize_t)r.authz.size()));
        } else {
            Py_INCREF(Py_None);
            PyList_SET_ITEM(authzs, (Py_ssize_t)i, Py_None);
        }
        if (r.kind == K_OTHER) {
            PyObject* m = Py_BuildValue("(s#s#y#)", r.method.data(), (Py_ssize_t)r.method.size(),
                                        r.target.data(), (Py_ssize_t)r.target.size(),
                                        r.headers_blob.data(), (Py_ssize_t)r.headers_blob.size());
            PyList_SET_ITEM(meta, (Py_ssize_t)i, m);
        } else {
            Py_INCREF(Py_None);
            PyList_SET_ITEM(meta, (Py_ssize_t)i, Py_None);
        }
    }
    return Py_BuildValue("(NNNNNN)", ids, kinds, bodies, users, authzs, meta);
}

PyObject* py_edge_complete(PyObject*, PyObject* args) {
    unsigned long long handle;
    PyObject *ids_b, *statuses_b, *bodies;
    if (!PyArg_ParseTuple(args, "KSSO", &handle, &ids_b, &statuses_b, &bodies)) return nullptr;
    Edge* e = (Edge*)(uintptr_t)handle;
    Py_ssize_t n = PyBytes_GET_SIZE(ids_b) / 8;
    if (PyBytes_GET_SIZE(statuses_b) < n * 2 || !PyList_Check(bodies) || PyList_GET_SIZE(bodies) < n) {
        PyErr_SetString(PyExc_ValueError, "edge_complete: length mismatch");
        return nullptr;
    }
    const uint64_t* idp = (const uint64_t*)PyBytes_AS_STRING(ids_b);
    const uint16_t* stp = (const uint16_t*)PyBytes_AS_STRING(statuses_b);
    // copy everything out under the GIL, then complete without it
    struct Item { uint64_t id; int status; std::string body; };
    std::vector<Item> items;
    items.reserve((size_t)n);
    for (Py_ssize_t i = 0; i < n; ++i) {
        PyObject* b = PyList_GET_ITEM(bodies, i);
        Item it;
        it.id = idp[i];
        it.status = stp[i];
        if (b != Py_None) {
            char* p;
            Py_ssize_t bn;
            if (PyBytes_AsStringAndSize(b, &p, &bn) != 0) return nullptr;
            it.body.assign(p, (size_t)bn);
        }
        items.push_back(std::move(it));
    }
    uint64_t done = 0;
    Py_BEGIN_ALLOW_THREADS;
    {
        std::vector<std::vector<uint32_t>> wake(e->threads.size());
        for (auto& it : items)
            if (complete_one(e, it.id, it.status, it.body.data(), it.body.size(), false, &wake))
                done++;
        for (size_t t = 0; t < wake.size(); ++t)
            e->threads[t]->wake_with_many(std::move(wake[t]));
    }
    Py_END_ALLOW_THREADS;
    return PyLong_FromUnsignedLongLong(done);
}

PyObject* py_edge_complete_raw(PyObject*, PyObject* args) {
    unsigned long long handle, id;
    Py_buffer raw;
    if (!PyArg_ParseTuple(args, "KKy*", &handle, &id, &raw)) return nullptr;
    Edge* e = (Edge*)(uintptr_t)handle;
    bool ok;
    Py_BEGIN_ALLOW_THREADS;
    ok = complete_one(e, id, 200, (const char*)raw.buf, (size_t)raw.len, true);
    Py_END_ALLOW_THREADS;
    PyBuffer_Release(&raw);
    return PyBool_FromLong(ok ? 1 : 0);
}

PyObject* py_edge_auth_put(PyObject*, PyObject* args) {
    unsigned long long handle;
    Py_buffer authz;
    PyObject* user;
    if (!PyArg_ParseTuple(args, "Ky*O", &handle, &authz, &user)) return nullptr;
    Edge* e = (Edge*)(uintptr_t)handle;
    std::string key((const char*)authz.buf, (size_t)authz.len);
    PyBuffer_Release(&authz);
    std::unique_lock<std::shared_mutex> g(e->auth_mu);
    if (user == Py_None) {
        Edge::AuthEntry& ent = e->auth[key];
        ent.user = nullptr;
        ent.neg_until = mono_s() + NEG_TTL;
    } else {
        const char* u = PyUnicode_AsUTF8(user);
        if (u == nullptr) return nullptr;
        e->interned.push_back(std::make_unique<std::string>(u));
        Edge::AuthEntry& ent = e->auth[key];
        ent.user = e->interned.back().get();
        ent.neg_until = 0.0;
    }
    Py_RETURN_NONE;
}

PyObject* py_edge_auth_clear(PyObject*, PyObject* args) {
    unsigned long long handle;
    if (!PyArg_ParseTuple(args, "K", &handle)) return nullptr;
    Edge* e = (Edge*)(uintptr_t)handle;
    std::unique_lock<std::shared_mutex> g(e->auth_mu);
    e->auth.clear();  // interned strings stay alive (queued Reqs point at them)
    Py_RETURN_NONE;
}

PyObject* py_edge_stats(PyObject*, PyObject* args) {
    unsigned long long handle;
    if (!PyArg_ParseTuple(args, "K", &handle)) return nullptr;
    Edge* e = (Edge*)(uintptr_t)handle;
    return Py_BuildValue(
        "{s:K,s:K,s:K,s:K,s:K,s:K,s:K,s:K,s:K,s:K}",
        "accepted", (unsigned long long)e->accepted.load(),
        "closed", (unsigned long long)e->closed_conns.load(),
        "hot", (unsigned long long)e->hot.load(),
        "cold", (unsigned long long)e->cold.load(),
        "direct_401", (unsigned long long)e->direct_401.load(),
        "direct_health", (unsigned long long)e->direct_health.load(),
        "responses", (unsigned long long)e->responses.load(),
        "bytes_in", (unsigned long long)e->bytes_in.load(),
        "bytes_out", (unsigned long long)e->bytes_out.load(),
        "parse_errors", (unsigned long long)e->parse_errors.load());
}

PyObject* py_edge_stop(PyObject*, PyObject* args) {
    unsigned long long handle;
    if (!PyArg_ParseTuple(args, "K", &handle)) return nullptr;
    Edge* e = (Edge*)(uintptr_t)handle;
    Py_BEGIN_ALLOW_THREADS;
    e->stop = true;
    e->qcv.notify_all();
    for (auto& t : e->threads)
        if (t->th.joinable()) t->th.join();
    Py_END_ALLOW_THREADS;
    delete e;
    Py_RETURN_NONE;
}

PyMethodDef edge_methods[] = {
    {"start", py_edge_start, METH_VARARGS,
     "start(port, nthreads, max_body, auth_required, backlog=4096) -> handle"},
    {"poll", py_edge_poll, METH_VARARGS,
     "poll(handle, wait_us, linger_us, max_n) -> (ids, kinds, bodies, users, authz, meta)"},
    {"complete", py_edge_complete, METH_VARARGS,
     "complete(handle, ids: bytes, statuses: bytes u16, bodies: list[bytes|None]) -> n_done"},
    {"complete_raw", py_edge_complete_raw, METH_VARARGS,
     "complete_raw(handle, id, raw_http_bytes) -> bool"},
    {"auth_put", py_edge_auth_put, METH_VARARGS, "auth_put(handle, authz: bytes, user: str|None)"},
    {"auth_clear", py_edge_auth_clear, METH_VARARGS, "auth_clear(handle)"},
    {"stats", py_edge_stats, METH_VARARGS, "stats(handle) -> dict"},
    {"stop", py_edge_stop, METH_VARARGS, "stop(handle)"},
    {nullptr, nullptr, 0, nullptr}};

struct PyModuleDef edge_module = {PyModuleDef_HEAD_INIT, "forge_edge",
                                  "native epoll HTTP edge for the GPU owner process",
                                  -1, edge_methods, nullptr, nullptr, nullptr, nullptr};

}  // namespace

PyMODINIT_FUNC PyInit_forge_edge(void) { return PyModule_Create(&edge_module); }
