// erasure_code_plugin.h — standalone mirror of the reference's plugin
// registry (src/erasure-code/ErasureCodePlugin.{h,cc}): dlopen of
// libec_<name>.so with RTLD_NOW, __erasure_code_version() gate,
// __erasure_code_init(name, dir) registration, and the factory() profile
// echo check. This is the drop-in boundary. In a real Ceph
// build the same shim sources compile against Ceph's own registry instead
// (INTEGRATION.md).
#pragma once

#include <functional>
#include <map>
#include <mutex>
#include <string>

#include "erasure_code.h"

// The harness's analogue of CEPH_GIT_NICE_VER: plugins built from this tree
// must return exactly this from __erasure_code_version()
// (ErasureCodePlugin.cc:162-171; mismatch => -EXDEV).
#define ECX_HARNESS_VERSION "ec-mi355x 0.1.0"

namespace ecx {

class ErasureCodePlugin {
 public:
  void *library = nullptr;
  virtual ~ErasureCodePlugin() = default;
  // ErasureCodePlugin.h:42-45
  virtual int factory(const std::string &directory,
                      ErasureCodeProfile &profile,
                      ErasureCodeInterfaceRef *erasure_code,
                      std::ostream *ss) = 0;
};

class ErasureCodePluginRegistry {
  std::mutex lock;
  std::map<std::string, ErasureCodePlugin *> plugins;
  int load(const std::string &plugin_name, const std::string &directory,
           ErasureCodePlugin **plugin, std::ostream *ss);

 public:
  bool disable_dlclose = false;
  static ErasureCodePluginRegistry &instance();
  ~ErasureCodePluginRegistry();
  int add(const std::string &name, ErasureCodePlugin *plugin);
  ErasureCodePlugin *get(const std::string &name);
  // ErasureCodePlugin.cc:104-132 — load if needed, construct, verify the
  // instance echoes the profile back exactly
  int factory(const std::string &plugin_name, const std::string &directory,
              ErasureCodeProfile &profile,
              ErasureCodeInterfaceRef *erasure_code, std::ostream *ss);
  int preload(const std::string &plugins_csv, const std::string &directory,
              std::ostream *ss);
};

}  // namespace ecx

// C entry points every plugin .so must export (ErasureCodePlugin.cc:33-34)
extern "C" {
typedef const char *(*ecx_plugin_version_fn)(void);
typedef int (*ecx_plugin_init_fn)(const char *, const char *);
}
