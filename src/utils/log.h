/* ucc_amd — logging. Reference parity: utils/debug/log.c (levels
 * FATAL..TRACE_POLL, env-configured per component); fresh implementation. */
#ifndef UCC_AMD_LOG_H_
#define UCC_AMD_LOG_H_

#include <cstdio>
#include <cstdarg>

namespace ucc {

enum class LogLevel : int {
    FATAL = 0,
    ERROR,
    WARN,
    INFO,
    DEBUG,
    TRACE,
    TRACE_POLL,
};

/* Global level, read once from UCC_LOG_LEVEL (fatal|error|warn|info|debug|
 * trace|trace_poll). Default WARN. */
LogLevel log_level();
void     log_set_level(LogLevel lvl);
void     log_print(LogLevel lvl, const char *component, const char *file,
                   int line, const char *fmt, ...)
    __attribute__((format(printf, 5, 6)));

#define UCC_LOG(lvl, comp, ...)                                               \
    do {                                                                      \
        if (static_cast<int>(lvl) <= static_cast<int>(::ucc::log_level())) {  \
            ::ucc::log_print(lvl, comp, __FILE__, __LINE__, __VA_ARGS__);     \
        }                                                                     \
    } while (0)

#define ucc_fatal(...) UCC_LOG(::ucc::LogLevel::FATAL, "core", __VA_ARGS__)
#define ucc_error(...) UCC_LOG(::ucc::LogLevel::ERROR, "core", __VA_ARGS__)
#define ucc_warn(...)  UCC_LOG(::ucc::LogLevel::WARN,  "core", __VA_ARGS__)
#define ucc_info(...)  UCC_LOG(::ucc::LogLevel::INFO,  "core", __VA_ARGS__)
#define ucc_debug(...) UCC_LOG(::ucc::LogLevel::DEBUG, "core", __VA_ARGS__)
#define ucc_trace(...) UCC_LOG(::ucc::LogLevel::TRACE, "core", __VA_ARGS__)

} // namespace ucc

#endif
