// plugin_fixture.cc — TEST FIXTURES for registry failure modes, compiled
// into several libec_*.so variants (see Makefile). Mirrors the reference's
// fixture plugins (src/test/erasure-code/ErasureCodePluginMissingVersion.cc,
// MissingEntryPoint, FailToInitialize, FailToRegister).
#include <cerrno>

#include "erasure_code_plugin.h"

extern "C" {

#ifndef FIXTURE_MISSING_VERSION
const char *__erasure_code_version() {
#ifdef FIXTURE_BAD_VERSION
  return "HelloWorld";
#else
  return ECX_HARNESS_VERSION;
#endif
}
#endif

#ifndef FIXTURE_MISSING_INIT
int __erasure_code_init(const char *plugin_name, const char *) {
#ifdef FIXTURE_FAIL_INIT
  (void)plugin_name;
  return -ESRCH;
#elif defined(FIXTURE_NO_REGISTER)
  (void)plugin_name;
  return 0;  // "forgets" to register => -EBADF from the registry
#else
  (void)plugin_name;
  return 0;
#endif
}
#endif
}
