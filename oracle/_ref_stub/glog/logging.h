// Minimal stand-in for glog (download-script-only thirdparty in the
// reference). Provides just the CHECK/DCHECK/LOG surface the included
// reference HEADERS use; aborts on failed CHECKs.
#pragma once
#include <cassert>
#include <cstdlib>
#include <iostream>
#include <sstream>

namespace google {
inline void InstallFailureSignalHandler() {}
}

struct _StubLogMsg {
    bool fatal;
    std::ostringstream os;
    _StubLogMsg(bool f) : fatal(f) {}
    ~_StubLogMsg() {
        if (fatal) { std::cerr << os.str() << std::endl; std::abort(); }
    }
    template <typename T>
    _StubLogMsg& operator<<(const T& v) { os << v; return *this; }
};
struct _StubVoidify {
    void operator&(_StubLogMsg&) {}
};

#define LOG(sev) _StubLogMsg(false)
#define VLOG(lvl) _StubLogMsg(false)
#define DLOG(sev) _StubLogMsg(false)
#define LOG_IF(sev, cond) _StubLogMsg(false)
#define LOG_EVERY_N(sev, n) _StubLogMsg(false)
#define CHECK(cond) ((cond) ? (void)0 : _StubVoidify() & _StubLogMsg(true) << "CHECK failed: " #cond)
#define CHECK_EQ(a, b) CHECK((a) == (b))
#define CHECK_NE(a, b) CHECK((a) != (b))
#define CHECK_LT(a, b) CHECK((a) < (b))
#define CHECK_LE(a, b) CHECK((a) <= (b))
#define CHECK_GT(a, b) CHECK((a) > (b))
#define CHECK_GE(a, b) CHECK((a) >= (b))
#define CHECK_NOTNULL(p) (p)
#define DCHECK(cond) CHECK(cond)
#define DCHECK_EQ(a, b) CHECK_EQ(a, b)
#define DCHECK_NE(a, b) CHECK_NE(a, b)
#define DCHECK_LT(a, b) CHECK_LT(a, b)
#define DCHECK_LE(a, b) CHECK_LE(a, b)
#define DCHECK_GT(a, b) CHECK_GT(a, b)
#define DCHECK_GE(a, b) CHECK_GE(a, b)
#define DCHECK_NOTNULL(p) (p)
