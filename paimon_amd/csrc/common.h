// Shared host-side utilities for libpaimon_hip.
#pragma once

#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <string>

namespace pmh {

// thread-local last error (exported via pmh_last_error()).
std::string &last_error();

inline void set_error(const char *fmt, ...) {
    char buf[1024];
    va_list ap;
    va_start(ap, fmt);
    vsnprintf(buf, sizeof(buf), fmt, ap);
    va_end(ap);
    last_error() = buf;
}

#define PMH_FAIL(...)            \
    do {                         \
        pmh::set_error(__VA_ARGS__); \
        return {};               \
    } while (0)

}  // namespace pmh
