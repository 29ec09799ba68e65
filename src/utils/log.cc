#include "log.h"

#include <cstring>
#include <cstdlib>
#include <ctime>
#include <unistd.h>
#include <sys/time.h>

namespace ucc {

static LogLevel parse_level_env()
{
    const char *e = getenv("UCC_LOG_LEVEL");
    if (!e) {
        return LogLevel::WARN;
    }
    struct {
        const char *name;
        LogLevel    lvl;
    } tbl[] = {
        {"fatal", LogLevel::FATAL}, {"error", LogLevel::ERROR},
        {"warn", LogLevel::WARN},   {"info", LogLevel::INFO},
        {"debug", LogLevel::DEBUG}, {"trace", LogLevel::TRACE},
        {"trace_poll", LogLevel::TRACE_POLL},
    };
    for (auto &t : tbl) {
        if (!strcasecmp(e, t.name)) {
            return t.lvl;
        }
    }
    return LogLevel::WARN;
}

static LogLevel g_level = parse_level_env();

LogLevel log_level() { return g_level; }
void     log_set_level(LogLevel lvl) { g_level = lvl; }

static const char *level_name(LogLevel l)
{
    switch (l) {
    case LogLevel::FATAL:      return "FATAL";
    case LogLevel::ERROR:      return "ERROR";
    case LogLevel::WARN:       return "WARN";
    case LogLevel::INFO:       return "INFO";
    case LogLevel::DEBUG:      return "DEBUG";
    case LogLevel::TRACE:      return "TRACE";
    case LogLevel::TRACE_POLL: return "POLL";
    }
    return "?";
}

void log_print(LogLevel lvl, const char *component, const char *file, int line,
               const char *fmt, ...)
{
    char    msg[2048];
    va_list ap;
    va_start(ap, fmt);
    vsnprintf(msg, sizeof(msg), fmt, ap);
    va_end(ap);

    struct timeval tv;
    gettimeofday(&tv, nullptr);
    const char *base = strrchr(file, '/');
    base             = base ? base + 1 : file;
    fprintf(stderr, "[%ld.%06ld] [%d] %-5s %s %s:%d %s\n", (long)tv.tv_sec,
            (long)tv.tv_usec, (int)getpid(), level_name(lvl), component, base,
            line, msg);
    if (lvl == LogLevel::FATAL) {
        abort();
    }
}

} // namespace ucc
