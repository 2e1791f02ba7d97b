// Persistent worker pool for the threaded host passes.
//
// Every hot C++ pass (envelope parse, decide, upstream batch, finalize,
// rewrite, post) previously spawned and joined 4-8 std::threads PER CALL
// — ~15 µs of pthread_create per thread adds up to ~0.5 ms per batch
// across the six passes. This pool keeps the workers alive; a call
// enqueues its range tasks and the calling thread (which holds no GIL —
// ctypes releases it) runs task 0 itself, then waits on a per-call
// counter. Multiple batches may call concurrently: tasks interleave in
// one queue and each call waits only on its own completion counter.
// Covered by the ASan/TSan driver (tests/test_sanitizers.py) which runs
// the passes from several outer threads at once.

#include <condition_variable>
#include <deque>
#include <mutex>
#include <thread>
#include <vector>

namespace {

struct Task {
    void (*fn)(int, void*);
    void* ctx;
    int index;
    struct Call* call;
};

struct Call {
    std::mutex mu;
    std::condition_variable cv;
    int remaining;
};

struct Pool {
    std::mutex mu;
    std::condition_variable cv;
    std::deque<Task> q;
    std::vector<std::thread> workers;

    explicit Pool(int n) {
        for (int i = 0; i < n; ++i)
            workers.emplace_back([this] { loop(); });
    }

    void loop() {
        for (;;) {
            Task t;
            {
                std::unique_lock<std::mutex> lk(mu);
                cv.wait(lk, [&] { return !q.empty(); });
                t = q.front();
                q.pop_front();
            }
            t.fn(t.index, t.ctx);
            {
                std::lock_guard<std::mutex> lk(t.call->mu);
                if (--t.call->remaining == 0) t.call->cv.notify_one();
            }
        }
    }
};

Pool* pool() {
    // leaked on purpose: joining detached-forever workers at static
    // destruction would hang interpreter shutdown
    static Pool* p = new Pool(14);
    return p;
}

}  // namespace

// Run fn(i, ctx) for i in [0, n); returns when ALL have finished. The
// calling thread runs task 0 (and only that), so a single-task call
// never touches the pool.
extern "C" void forge_parallel_for(int n, void (*fn)(int, void*), void* ctx) {
    if (n <= 0) return;
    if (n == 1) {
        fn(0, ctx);
        return;
    }
    Call call;
    call.remaining = n - 1;
    Pool* p = pool();
    {
        std::lock_guard<std::mutex> lk(p->mu);
        for (int i = 1; i < n; ++i)
            p->q.push_back(Task{fn, ctx, i, &call});
    }
    p->cv.notify_all();
    fn(0, ctx);
    std::unique_lock<std::mutex> lk(call.mu);
    call.cv.wait(lk, [&] { return call.remaining == 0; });
}
