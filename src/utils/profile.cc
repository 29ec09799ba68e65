/* See profile.h. */
#include "profile.h"

#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <ctime>
#include <string>

#include <unistd.h>

#include "config.h"

namespace ucc {
namespace prof {

struct Rec {
    double      t;
    Ev          type;
    const char *name;
    uint64_t    id;
};

constexpr size_t kCap = 1 << 18; /* 256k records, ~8 MB */

static Rec                 *g_ring  = nullptr;
static std::atomic<size_t>  g_n{0};
static bool                 g_on    = false;
static std::string          g_file;

static double now_s()
{
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return ts.tv_sec + 1e-9 * ts.tv_nsec;
}

static bool init_once()
{
    static bool inited = [] {
        auto &cfg = Config::instance();
        cfg.declare("", "PROFILE_MODE", "off",
                    "request-event profiling: off | log");
        cfg.declare("", "PROFILE_FILE", "ucc_profile.log",
                    "profile dump path (%p -> pid)");
        std::string mode = cfg.get("", "PROFILE_MODE", "off");
        if (mode == "log" || mode == "on" || mode == "1") {
            g_on   = true;
            g_file = cfg.get("", "PROFILE_FILE", "ucc_profile.log");
            size_t p = g_file.find("%p");
            if (p != std::string::npos) {
                g_file.replace(p, 2, std::to_string((int)getpid()));
            }
            g_ring = (Rec *)calloc(kCap, sizeof(Rec));
            atexit([] { dump(); });
        }
        return true;
    }();
    return inited;
}

bool enabled()
{
    init_once();
    return g_on;
}

void record(Ev type, const char *name, uint64_t req_id)
{
    if (!g_ring) {
        return;
    }
    size_t i        = g_n.fetch_add(1, std::memory_order_relaxed);
    Rec   &r        = g_ring[i % kCap];
    r.t             = now_s();
    r.type          = type;
    r.name          = name;
    r.id            = req_id;
}

static const char *type_name(Ev t)
{
    switch (t) {
    case Ev::REQUEST_NEW: return "new";
    case Ev::REQUEST_EVENT: return "event";
    case Ev::REQUEST_FREE: return "free";
    case Ev::SCOPE_BEGIN: return "begin";
    case Ev::SCOPE_END: return "end";
    }
    return "?";
}

void dump()
{
    if (!g_ring) {
        return;
    }
    size_t n = g_n.load(std::memory_order_relaxed);
    if (n == 0) {
        return;
    }
    FILE *f = fopen(g_file.c_str(), "w");
    if (!f) {
        return;
    }
    fprintf(f, "# ucc_amd profile pid=%d records=%zu (cap %zu)\n",
            (int)getpid(), n, kCap);
    size_t start = n > kCap ? n - kCap : 0;
    for (size_t i = start; i < n; i++) {
        Rec &r = g_ring[i % kCap];
        fprintf(f, "%.9f %s %s req=%llu\n", r.t, type_name(r.type),
                r.name ? r.name : "?", (unsigned long long)r.id);
    }
    fclose(f);
    g_n.store(0, std::memory_order_relaxed);
}

} // namespace prof
} // namespace ucc
