// ec_benchmark.cc — standalone mirror of the reference's
// ceph_erasure_code_benchmark (src/test/erasure-code/
// ceph_erasure_code_benchmark.cc): same flags (--plugin/-p, --size/-s,
// --iterations/-i, --workload/-w, --erasures/-e, --erased,
// --erasures-generation/-E, --parameter/-P) plus --directory (the
// erasure_code_dir conf option, global.yaml.in:470), and the same
// two-column "seconds \t total-KiB-of-input" output (benchmark.cc:193,324).
// Deviations: input is seeded-random bytes instead of constant 'X'
// (constant fill can mask a broken codec), recorded on stderr.
#include <getopt.h>

#include <chrono>
#include <cstring>
#include <iostream>
#include <random>
#include <sstream>
#include <vector>

#include "erasure_code_plugin.h"

using namespace ecx;

struct Bench {
  int in_size = 80 * 1024 * 1024;  // benchmark.cc:55
  int max_iterations = 100;
  std::string plugin = "mi355x";
  std::string workload = "encode";
  std::string directory = ".";
  int erasures = 1;
  bool exhaustive = false;
  std::vector<int> erased;
  ErasureCodeProfile profile;
  bool verbose = false;
  int k = 0, m = 0;
  uint64_t seed = 0xEC;

  int setup(int argc, char **argv);
  int run();
  int encode();
  int decode();
  int flags();
  int decode_erasures(const shard_id_map<buffer> &all,
                      const shard_id_map<buffer> &chunks, int shard,
                      unsigned want_erasures,
                      ErasureCodeInterfaceRef erasure_code);
  buffer make_input() const;
};

int Bench::setup(int argc, char **argv) {
  static option longopts[] = {
      {"help", no_argument, nullptr, 'h'},
      {"verbose", no_argument, nullptr, 'v'},
      {"size", required_argument, nullptr, 's'},
      {"iterations", required_argument, nullptr, 'i'},
      {"plugin", required_argument, nullptr, 'p'},
      {"workload", required_argument, nullptr, 'w'},
      {"erasures", required_argument, nullptr, 'e'},
      {"erased", required_argument, nullptr, 1000},
      {"erasures-generation", required_argument, nullptr, 'E'},
      {"parameter", required_argument, nullptr, 'P'},
      {"directory", required_argument, nullptr, 'd'},
      {"seed", required_argument, nullptr, 1001},
      {"flags", no_argument, nullptr, 1002},
      {nullptr, 0, nullptr, 0}};
  int c;
  while ((c = getopt_long(argc, argv, "hvs:i:p:w:e:E:P:d:", longopts,
                          nullptr)) != -1) {
    switch (c) {
      case 'h':
        std::cout
            << "usage: ec_benchmark [-p plugin] [-s size] [-i iterations]\n"
               "  [-w encode|decode] [-e erasures] [--erased N ...]\n"
               "  [-E random|exhaustive] [-P k=v ...] [-d plugin-dir]\n";
        return 1;
      case 'v': verbose = true; break;
      case 's': in_size = atoi(optarg); break;
      case 'i': max_iterations = atoi(optarg); break;
      case 'p': plugin = optarg; break;
      case 'w': workload = optarg; break;
      case 'e': erasures = atoi(optarg); break;
      case 'E': exhaustive = std::string(optarg) == "exhaustive"; break;
      case 'd': directory = optarg; break;
      case 1000: erased.push_back(atoi(optarg)); break;
      case 1001: seed = strtoull(optarg, nullptr, 0); break;
      case 1002: workload = "flags"; break;
      case 'P': {
        std::string s(optarg);
        auto eq = s.find('=');
        if (eq == std::string::npos) {
          std::cerr << "--parameter " << s << " ignored (no =)\n";
        } else {
          profile[s.substr(0, eq)] = s.substr(eq + 1);
        }
        break;
      }
      default: return -EINVAL;
    }
  }
  // k/m usually come from the profile (benchmark.cc:135-141); for plugins
  // whose profile omits them (LRC with explicit layers) they are derived
  // from the constructed codec after factory().
  try {
    k = std::stoi(profile.at("k"));
    m = std::stoi(profile.at("m"));
  } catch (const std::exception &) {
    k = m = 0;
  }
  return 0;
}

buffer Bench::make_input() const {
  buffer in = buffer::create_aligned(in_size, ErasureCode::SIMD_ALIGN);
  std::mt19937_64 rng(seed);
  uint64_t *p = (uint64_t *)in.c_str();
  size_t words = in_size / 8;
  for (size_t i = 0; i < words; i++) p[i] = rng();
  for (size_t i = words * 8; i < (size_t)in_size; i++) in.c_str()[i] = 'X';
  return in;
}

// --flags: factory the plugin and print its claimed optimization flags
// (names per ErasureCodeInterface.h:694-709) so conformance tests can
// assert claims == verified behaviour.
int Bench::flags() {
  auto &instance = ErasureCodePluginRegistry::instance();
  ErasureCodeInterfaceRef erasure_code;
  std::stringstream messages;
  int code =
      instance.factory(plugin, directory, profile, &erasure_code, &messages);
  if (code) {
    std::cerr << messages.str() << std::endl;
    return code;
  }
  auto f = erasure_code->get_supported_optimizations();
  static const std::pair<uint64_t, const char *> names[] = {
      {1 << 0, "partialread"},    {1 << 1, "partialwrite"},
      {1 << 2, "zeroinout"},      {1 << 3, "zeropadding"},
      {1 << 4, "paritydelta"},    {1 << 5, "requiresubchunks"},
      {1 << 6, "optimizedsupport"}, {1 << 7, "crcencodedecode"},
      {1 << 8, "directreads"}};
  bool first = true;
  for (auto &[bit, name] : names) {
    if (f & bit) {
      std::cout << (first ? "" : ",") << name;
      first = false;
    }
  }
  std::cout << std::endl;
  return 0;
}

int Bench::run() {
  ErasureCodePluginRegistry::instance().disable_dlclose = true;
  if (workload == "encode") return encode();
  if (workload == "flags") return flags();
  return decode();
}

int Bench::encode() {
  // mirror of ErasureCodeBench::encode (benchmark.cc:165-195)
  auto &instance = ErasureCodePluginRegistry::instance();
  ErasureCodeInterfaceRef erasure_code;
  std::stringstream messages;
  int code =
      instance.factory(plugin, directory, profile, &erasure_code, &messages);
  if (code) {
    std::cerr << messages.str() << std::endl;
    return code;
  }
  if (k <= 0) {
    k = erasure_code->get_data_chunk_count();
    m = erasure_code->get_chunk_count() - k;
  }
  buffer in = make_input();
  shard_id_set want_to_encode;
  for (int i = 0; i < (int)erasure_code->get_chunk_count(); i++)
    want_to_encode.insert(i);
  auto begin = std::chrono::steady_clock::now();
  for (int i = 0; i < max_iterations; i++) {
    shard_id_map<buffer> encoded(erasure_code->get_chunk_count());
    code = erasure_code->encode(want_to_encode, in, &encoded);
    if (code) return code;
  }
  auto end = std::chrono::steady_clock::now();
  double secs = std::chrono::duration<double>(end - begin).count();
  std::cout << secs << "\t" << ((uint64_t)max_iterations * (in_size / 1024))
            << std::endl;
  return 0;
}

int Bench::decode_erasures(const shard_id_map<buffer> &all,
                           const shard_id_map<buffer> &chunks, int shard,
                           unsigned want_erasures,
                           ErasureCodeInterfaceRef erasure_code) {
  // mirror of benchmark.cc:211-258 (recursive exhaustive erasure + verify)
  if (want_erasures == 0) {
    shard_id_set want_to_read;
    for (int c = 0; c < (int)erasure_code->get_chunk_count(); c++)
      if (!chunks.contains(c)) want_to_read.insert(c);
    shard_id_map<buffer> decoded(erasure_code->get_chunk_count());
    int code = erasure_code->decode(want_to_read, chunks, &decoded, 0);
    if (code) return code;
    for (auto &&s : want_to_read) {
      const buffer &a = all.at(s);
      buffer &b = decoded[s];
      if (a.length() != b.length() ||
          std::memcmp(a.c_str(), b.c_str(), a.length()) != 0) {
        std::cerr << "chunk " << (int)s << " recovered content mismatch\n";
        return -1;
      }
    }
    return 0;
  }
  for (; shard < (int)erasure_code->get_chunk_count(); shard++) {
    shard_id_map<buffer> one_less = chunks;
    one_less.erase(shard);
    int code = decode_erasures(all, one_less, shard + 1, want_erasures - 1,
                               erasure_code);
    if (code) return code;
  }
  return 0;
}

int Bench::decode() {
  // mirror of ErasureCodeBench::decode (benchmark.cc:260-326)
  auto &instance = ErasureCodePluginRegistry::instance();
  ErasureCodeInterfaceRef erasure_code;
  std::stringstream messages;
  int code =
      instance.factory(plugin, directory, profile, &erasure_code, &messages);
  if (code) {
    std::cerr << messages.str() << std::endl;
    return code;
  }
  if (k <= 0) {
    k = erasure_code->get_data_chunk_count();
    m = erasure_code->get_chunk_count() - k;
  }
  buffer in = make_input();
  shard_id_set want_all;
  for (int i = 0; i < (int)erasure_code->get_chunk_count(); i++)
    want_all.insert(i);
  shard_id_map<buffer> encoded(erasure_code->get_chunk_count());
  code = erasure_code->encode(want_all, in, &encoded);
  if (code) return code;

  if (!erased.empty())
    for (int e : erased) encoded.erase(e);

  std::mt19937 rng((uint32_t)seed);
  auto begin = std::chrono::steady_clock::now();
  for (int i = 0; i < max_iterations; i++) {
    if (exhaustive) {
      code = decode_erasures(encoded, encoded, 0, erasures, erasure_code);
      if (code) return code;
    } else if (!erased.empty()) {
      shard_id_map<buffer> decoded(erasure_code->get_chunk_count());
      code = erasure_code->decode(want_all, encoded, &decoded, 0);
      if (code) return code;
    } else {
      shard_id_map<buffer> chunks = encoded;
      for (int j = 0; j < erasures; j++) {
        int e;
        do {
          e = rng() % (k + m);
        } while (!chunks.contains(e));
        chunks.erase(e);
      }
      shard_id_map<buffer> decoded(erasure_code->get_chunk_count());
      code = erasure_code->decode(want_all, chunks, &decoded, 0);
      if (code) return code;
    }
  }
  auto end = std::chrono::steady_clock::now();
  double secs = std::chrono::duration<double>(end - begin).count();
  std::cout << secs << "\t" << ((uint64_t)max_iterations * (in_size / 1024))
            << std::endl;
  return 0;
}

int main(int argc, char **argv) {
  Bench bench;
  int err = bench.setup(argc, argv);
  if (err) return err > 0 ? 0 : err;
  return bench.run();
}
