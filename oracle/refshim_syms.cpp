// Minimal stand-ins for Arcadia runtime leaves referenced by the reference
// translation units the byte-fixture harness compiles (assert/abort
// plumbing, allocator poison, logging). Signatures come from the real
// headers; none of this affects the bit-pack byte layout under test.
#include <util/generic/singleton.h>
#include <util/generic/strbuf.h>
#include <util/stream/output.h>
#include <util/system/atexit.h>
#include <cstdarg>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <typeinfo>

void AtExit(TAtExitFunc fn, void* arg, size_t) { (void)fn; (void)arg; }
void FormatBackTrace(IOutputStream*) {}

namespace NPrivate {
void FillWithTrash(void* p, size_t len) { memset(p, 0, len); }
void LockRecursive(std::atomic<size_t>&) noexcept {}
void UnlockRecursive(std::atomic<size_t>&) noexcept {}
[[noreturn]] void Panic(const TStaticBuf&, int, const char*, const char*, const char*, ...) noexcept { abort(); }
IOutputStream& StdErrStream() noexcept;
} // namespace NPrivate

namespace NSystemInfo { size_t GetPageSize() { return 4096; } }

namespace NStringSplitPrivate { extern const char SPLITTER_EMPTY_SENTINEL[]; const char SPLITTER_EMPTY_SENTINEL[1] = {0}; }

namespace NYT {
namespace NDetail {
[[noreturn]] void AbortOnOom() { abort(); }
[[noreturn]] void AssertTrapImpl(TStringBuf, TStringBuf, TStringBuf, TStringBuf, int, TStringBuf) { abort(); }
} // namespace NDetail
namespace NLogging {
struct TLoggingContext;
} // namespace NLogging
} // namespace NYT
#include <library/cpp/yt/error/error_code.h>
namespace NYT {
TErrorCodeRegistry* TErrorCodeRegistry::Get() { static TErrorCodeRegistry r; return &r; }
void TErrorCodeRegistry::RegisterErrorCode(int, const TErrorCodeInfo&) {}
std::string TErrorCodeRegistry::ParseNamespace(const std::type_info&) { return ""; }
} // namespace NYT

// --- backtrace stand-ins (util/system/backtrace.h API) ---
#include <util/system/backtrace.h>
TBackTrace::TBackTrace() : Size(0) {}
void TBackTrace::Capture() {}
void TBackTrace::PrintTo(IOutputStream&) const {}
TString TBackTrace::PrintToString() const { return TString(); }
size_t TBackTrace::size() const { return 0; }
const void* const* TBackTrace::data() const { return nullptr; }
TBackTrace TBackTrace::FromCurrentException() { return TBackTrace(); }
