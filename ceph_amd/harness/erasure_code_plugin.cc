// erasure_code_plugin.cc — see erasure_code_plugin.h. Mirrors
// src/erasure-code/ErasureCodePlugin.cc load/factory semantics including
// its error codes (-EIO dlopen failure, -EXDEV version mismatch, -ENOENT
// missing init, -EBADF init-didn't-register, -EEXIST double add).
#include "erasure_code_plugin.h"

#include <dlfcn.h>

#include <cerrno>
#include <ostream>
#include <sstream>

namespace ecx {

ErasureCodePluginRegistry &ErasureCodePluginRegistry::instance() {
  static ErasureCodePluginRegistry singleton;
  return singleton;
}

ErasureCodePluginRegistry::~ErasureCodePluginRegistry() {
  for (auto &[name, plugin] : plugins) {
    void *library = plugin->library;
    delete plugin;
    if (library && !disable_dlclose) dlclose(library);
  }
}

int ErasureCodePluginRegistry::add(const std::string &name,
                                   ErasureCodePlugin *plugin) {
  if (plugins.count(name)) return -EEXIST;
  plugins[name] = plugin;
  return 0;
}

ErasureCodePlugin *ErasureCodePluginRegistry::get(const std::string &name) {
  auto it = plugins.find(name);
  return it == plugins.end() ? nullptr : it->second;
}

int ErasureCodePluginRegistry::factory(const std::string &plugin_name,
                                       const std::string &directory,
                                       ErasureCodeProfile &profile,
                                       ErasureCodeInterfaceRef *erasure_code,
                                       std::ostream *ss) {
  ErasureCodePlugin *plugin;
  {
    std::lock_guard<std::mutex> l{lock};
    plugin = get(plugin_name);
    if (plugin == nullptr) {
      int r = load(plugin_name, directory, &plugin, ss);
      if (r != 0) return r;
    }
  }
  int r = plugin->factory(directory, profile, erasure_code, ss);
  if (r) return r;
  // profile echo gate (ErasureCodePlugin.cc:126-130)
  if (profile != (*erasure_code)->get_profile()) {
    if (ss) {
      *ss << "factory: profile {";
      for (auto &[k, v] : profile) *ss << k << "=" << v << ",";
      *ss << "} != get_profile() {";
      for (auto &[k, v] : (*erasure_code)->get_profile())
        *ss << k << "=" << v << ",";
      *ss << "}\n";
    }
    return -EINVAL;
  }
  return 0;
}

static const char *an_older_version() { return "an older version"; }

int ErasureCodePluginRegistry::load(const std::string &plugin_name,
                                    const std::string &directory,
                                    ErasureCodePlugin **plugin,
                                    std::ostream *ss) {
  // ErasureCodePlugin.cc:138-206
  std::string fname = directory + "/libec_" + plugin_name + ".so";
  void *library = dlopen(fname.c_str(), RTLD_NOW);
  if (!library) {
    if (ss) *ss << "load dlopen(" << fname << "): " << dlerror();
    return -EIO;
  }

  auto version =
      (ecx_plugin_version_fn)dlsym(library, "__erasure_code_version");
  if (version == nullptr) version = an_older_version;
  if (std::string(version()) != ECX_HARNESS_VERSION) {
    if (ss)
      *ss << "expected plugin " << fname << " version " << ECX_HARNESS_VERSION
          << " but it claims to be " << version() << " instead";
    dlclose(library);
    return -EXDEV;
  }

  auto init = (ecx_plugin_init_fn)dlsym(library, "__erasure_code_init");
  if (init) {
    int r = init(plugin_name.c_str(), directory.c_str());
    if (r != 0) {
      if (ss)
        *ss << "erasure_code_init(" << plugin_name << "," << directory
            << "): " << r;
      dlclose(library);
      return r;
    }
  } else {
    if (ss)
      *ss << "load dlsym(" << fname << ", __erasure_code_init): "
          << dlerror();
    dlclose(library);
    return -ENOENT;
  }

  *plugin = get(plugin_name);
  if (*plugin == nullptr) {
    if (ss)
      *ss << "load __erasure_code_init() did not register " << plugin_name;
    dlclose(library);
    return -EBADF;
  }
  (*plugin)->library = library;
  return 0;
}

int ErasureCodePluginRegistry::preload(const std::string &plugins_csv,
                                       const std::string &directory,
                                       std::ostream *ss) {
  std::lock_guard<std::mutex> l{lock};
  std::stringstream s(plugins_csv);
  std::string name;
  while (std::getline(s, name, ',')) {
    if (name.empty()) continue;
    ErasureCodePlugin *plugin;
    int r = load(name, directory, &plugin, ss);
    if (r) return r;
  }
  return 0;
}

}  // namespace ecx
