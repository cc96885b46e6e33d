// C-ABI scaffolding: thread-local error string + build info.
#include "vh_common.h"

#include <cstdarg>

namespace {
thread_local char g_err[1024] = "";
}

void vh_set_error(const char* fmt, ...) {
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(g_err, sizeof(g_err), fmt, ap);
  va_end(ap);
}

extern "C" const char* vh_last_error(void) { return g_err; }

extern "C" const char* vh_build_info(void) {
  return "veomni_hip gfx950 v0.1 (" __DATE__ " " __TIME__ ")";
}
