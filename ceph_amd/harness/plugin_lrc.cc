// plugin_lrc.cc — libec_lrc.so: locally-repairable codes as layered
// composition over registry sub-plugins, mirroring the reference's LRC
// plugin (src/erasure-code/lrc/ErasureCodeLrc.cc): parse_kml expansion
// (:292-395), layer chunk-map parsing and sub-plugin instantiation via the
// registry (:140-249; default sub-plugin here is "mi355x" instead of the
// reference's "isa", override with profile key "lrc-default-plugin" — e.g.
// "oracle" for CPU tests), layered encode (:952-1005), reverse-layer decode
// (:1093-1160), and the locality-aware minimum_to_decode (:578-700 case
// 1/2 logic). All GF compute happens in the sub-plugins (GPU for mi355x).
//
// NOTE on BASELINE configs[3] "LRC k=8 m=3 l=4": the reference's own
// parse_kml REJECTS it ((k+m) % l != 0, ERROR_LRC_K_M_MODULO); the bench
// uses the nearest valid shape (k=9 m=3 l=4 or the doc example k=4 m=2
// l=3) and says so.
#include <algorithm>
#include <cerrno>
#include <cstring>
#include <memory>
#include <ostream>
#include <sstream>
#include <vector>

#include "erasure_code_plugin.h"

using namespace ecx;

namespace {

// minimal JSON-ish parser for the layers array: [ [ "map", "profile" ], .. ]
// where the second element is a string of space-separated k=v pairs or a
// flat {"k":"v"} object — covers everything parse_kml generates plus
// hand-written profiles of that shape.
struct LayersParser {
  const std::string &s;
  size_t i = 0;
  explicit LayersParser(const std::string &str) : s(str) {}
  void ws() { while (i < s.size() && isspace((unsigned char)s[i])) i++; }
  bool eat(char c) {
    ws();
    if (i < s.size() && s[i] == c) { i++; return true; }
    return false;
  }
  bool str(std::string *out) {
    ws();
    if (i >= s.size() || s[i] != '"') return false;
    i++;
    out->clear();
    while (i < s.size() && s[i] != '"') out->push_back(s[i++]);
    if (i >= s.size()) return false;
    i++;
    return true;
  }
  // parse {"k": "v", ...} into profile
  bool obj(ErasureCodeProfile *p) {
    if (!eat('{')) return false;
    ws();
    if (eat('}')) return true;
    do {
      std::string k, v;
      if (!str(&k) || !eat(':') || !str(&v)) return false;
      (*p)[k] = v;
    } while (eat(','));
    return eat('}');
  }
};

static int kv_string_to_profile(const std::string &in,
                                ErasureCodeProfile *p) {
  std::stringstream ss(in);
  std::string tok;
  while (ss >> tok) {
    auto eq = tok.find('=');
    if (eq == std::string::npos) return -EINVAL;
    (*p)[tok.substr(0, eq)] = tok.substr(eq + 1);
  }
  return 0;
}

class ErasureCodeLrc final : public ErasureCode {
 public:
  struct Layer {
    std::string chunks_map;
    ErasureCodeProfile profile;
    ErasureCodeInterfaceRef erasure_code;
    std::vector<int> data, coding, chunks;
    shard_id_set chunk_set;
  };
  std::vector<Layer> layers;
  std::string directory;
  std::string default_plugin = "mi355x";
  unsigned int chunk_count_ = 0, data_chunk_count_ = 0;

  explicit ErasureCodeLrc(std::string dir) : directory(std::move(dir)) {}

  unsigned int get_chunk_count() const override { return chunk_count_; }
  unsigned int get_data_chunk_count() const override {
    return data_chunk_count_;
  }
  plugin_flags get_supported_optimizations() const override {
    // ErasureCodeLrc.h:108-112
    return FLAG_EC_PLUGIN_PARTIAL_READ_OPTIMIZATION |
           FLAG_EC_PLUGIN_PARTIAL_WRITE_OPTIMIZATION |
           FLAG_EC_PLUGIN_ZERO_INPUT_ZERO_OUTPUT_OPTIMIZATION;
  }
  unsigned int get_chunk_size(unsigned int stripe_width) const override {
    return layers.front().erasure_code->get_chunk_size(stripe_width);
  }

  // parse_kml expansion (ErasureCodeLrc.cc:292-395)
  int parse_kml(ErasureCodeProfile &profile, std::ostream *ss) {
    int err = 0;
    int k = -1, m = -1, l = -1;
    err |= to_int("k", profile, &k, "-1", ss);
    err |= to_int("m", profile, &m, "-1", ss);
    err |= to_int("l", profile, &l, "-1", ss);
    if (k == -1 && m == -1 && l == -1) return err;
    if (k == -1 || m == -1 || l == -1) {
      if (ss) *ss << "all of k, m, l must be set or none\n";
      return -EINVAL;
    }
    for (const char *g : {"mapping", "layers"}) {
      if (profile.count(g)) {
        if (ss) *ss << g << " cannot be set when k, m, l are set\n";
        return -EINVAL;
      }
    }
    if (l == 0 || (k + m) % l) {
      if (ss) *ss << "k + m must be a multiple of l\n";
      return -EINVAL;
    }
    int groups = (k + m) / l;
    if (k % groups || m % groups) {
      if (ss) *ss << "k and m must be multiples of (k + m) / l\n";
      return -EINVAL;
    }
    std::string mapping;
    for (int i = 0; i < groups; i++)
      mapping += std::string(k / groups, 'D') +
                 std::string(m / groups, '_') + "_";
    profile["mapping"] = mapping;

    std::string layers_str = "[ ";
    layers_str += " [ \"";
    for (int i = 0; i < groups; i++)
      layers_str += std::string(k / groups, 'D') +
                    std::string(m / groups, 'c') + "_";
    layers_str += "\", \"\" ],";
    for (int i = 0; i < groups; i++) {
      layers_str += " [ \"";
      for (int j = 0; j < groups; j++)
        layers_str += (i == j) ? std::string(l, 'D') + "c"
                               : std::string(l + 1, '_');
      layers_str += "\", \"\" ],";
    }
    profile["layers"] = layers_str + "]";
    return err;
  }

  int layers_parse(const std::string &desc, std::ostream *ss) {
    LayersParser p(desc);
    if (!p.eat('[')) {
      if (ss) *ss << "layers must be a JSON array: " << desc << "\n";
      return -EINVAL;
    }
    p.ws();
    while (true) {
      p.ws();
      if (p.eat(']')) break;
      if (!p.eat('[')) {
        if (ss) *ss << "layer entry must be an array: " << desc << "\n";
        return -EINVAL;
      }
      Layer layer;
      if (!p.str(&layer.chunks_map)) return -EINVAL;
      if (p.eat(',')) {
        p.ws();
        if (p.i < p.s.size() && p.s[p.i] == '{') {
          if (!p.obj(&layer.profile)) return -EINVAL;
        } else {
          std::string prof_str;
          if (!p.str(&prof_str)) return -EINVAL;
          if (kv_string_to_profile(prof_str, &layer.profile)) return -EINVAL;
        }
      }
      if (!p.eat(']')) return -EINVAL;
      p.eat(',');
      layers.push_back(std::move(layer));
    }
    return 0;
  }

  int layers_init(std::ostream *ss) {
    // ErasureCodeLrc.cc:210-249 with default sub-plugin mi355x
    auto &registry = ErasureCodePluginRegistry::instance();
    for (auto &layer : layers) {
      int position = 0;
      for (char c : layer.chunks_map) {
        if (c == 'D') layer.data.push_back(position);
        if (c == 'c') layer.coding.push_back(position);
        if (c == 'c' || c == 'D') layer.chunk_set.insert(position);
        position++;
      }
      layer.chunks = layer.data;
      layer.chunks.insert(layer.chunks.end(), layer.coding.begin(),
                          layer.coding.end());
      if (!layer.profile.count("k"))
        layer.profile["k"] = std::to_string(layer.data.size());
      if (!layer.profile.count("m"))
        layer.profile["m"] = std::to_string(layer.coding.size());
      if (!layer.profile.count("plugin"))
        layer.profile["plugin"] = default_plugin;
      if (!layer.profile.count("technique"))
        layer.profile["technique"] = "reed_sol_van";
      int err = registry.factory(layer.profile["plugin"], directory,
                                 layer.profile, &layer.erasure_code, ss);
      if (err) return err;
    }
    return 0;
  }

  int init(ErasureCodeProfile &profile, std::ostream *ss) override {
    to_string("lrc-default-plugin", profile, &default_plugin, "mi355x", ss);
    int r = parse_kml(profile, ss);
    if (r) return r;
    r = ErasureCode::parse(profile, ss);  // mapping -> chunk_mapping
    if (r) return r;
    if (!profile.count("layers")) {
      if (ss) *ss << "the layers parameter is missing\n";
      return -EINVAL;
    }
    r = layers_parse(profile["layers"], ss);
    if (r) return r;
    r = layers_init(ss);
    if (r) return r;
    if (!profile.count("mapping")) {
      if (ss) *ss << "the mapping parameter is missing\n";
      return -EINVAL;
    }
    const std::string &mapping = profile["mapping"];
    data_chunk_count_ =
        std::count(mapping.begin(), mapping.end(), 'D');
    chunk_count_ = mapping.size();
    for (auto &layer : layers)
      if (layer.chunks_map.size() != chunk_count_) {
        if (ss)
          *ss << "layer map " << layer.chunks_map << " length != "
              << chunk_count_ << "\n";
        return -EINVAL;
      }
    // kml-generated params are not exposed (ErasureCodeLrc.cc:667-675)
    if (profile.count("l") && profile["l"] != "-1") {
      profile.erase("mapping");
      profile.erase("layers");
    }
    return ErasureCode::init(profile, ss);
  }

  int encode_chunks(const shard_id_map<buffer> &in,
                    shard_id_map<buffer> &out) override {
    // ErasureCodeLrc.cc:952-1005: walk layers top-down; each layer sees
    // reindexed chunks; buffers written by earlier layers (global parity)
    // feed later local layers through the out map.
    shard_id_set all_shards;
    for (auto &&[shard, b] : in) { (void)b; all_shards.insert(shard); }
    for (auto &&[shard, b] : out) { (void)b; all_shards.insert(shard); }

    unsigned top = layers.size();
    for (auto i = layers.rbegin(); i != layers.rend(); ++i) {
      --top;
      if (i->chunk_set.includes(all_shards)) break;
    }
    for (unsigned li = top; li < layers.size(); ++li) {
      Layer &layer = layers[li];
      shard_id_map<buffer> layer_in(get_chunk_count());
      shard_id_map<buffer> layer_out(get_chunk_count());
      int j = 0;
      for (int c : layer.chunks) {
        if (in.contains(c)) layer_in[j] = in.at(shard_id_t((int8_t)c));
        if (out.contains(c)) layer_out[j] = out.at(shard_id_t((int8_t)c));
        ++j;
      }
      int err = layer.erasure_code->encode_chunks(layer_in, layer_out);
      if (err) return err;
    }
    return 0;
  }

  int decode_chunks(const shard_id_set &want_to_read,
                    shard_id_map<buffer> &in,
                    shard_id_map<buffer> &out) override {
    // ErasureCodeLrc.cc:1093-1160: reverse layer order (locals first)
    shard_id_set erasures;
    for (auto &&[shard, b] : out) { (void)b; erasures.insert(shard); }

    // seed with the wanted erasures so that a decode where EVERY layer is
    // skipped (e.g. a raw caller supplied too few buffers) reports -EIO
    // instead of silently returning unrecovered chunks
    shard_id_set want_to_read_erasures;
    for (auto s : want_to_read)
      if (erasures.contains(s)) want_to_read_erasures.insert(s);
    for (auto layer = layers.rbegin(); layer != layers.rend(); ++layer) {
      shard_id_set layer_erasures;
      for (auto s : layer->chunk_set)
        if (erasures.contains(s)) layer_erasures.insert(s);
      if (layer_erasures.size() >
              layer->erasure_code->get_coding_chunk_count() ||
          layer_erasures.empty())
        continue;
      // A raw decode_chunks caller (not ErasureCode::_decode, which
      // invents buffers for every chunk) may omit buffers for chunks it
      // considers unavailable; a layer whose survivors are not all backed
      // by a buffer cannot run — skip it instead of crashing on .at()
      bool layer_backed = true;
      for (int c : layer->chunks) {
        shard_id_t cs((int8_t)c);
        if (!erasures.contains(cs) && !in.contains(cs) && !out.contains(cs)) {
          layer_backed = false;
          break;
        }
      }
      if (!layer_backed) continue;
      shard_id_set layer_want;
      shard_id_map<buffer> layer_in(get_chunk_count());
      shard_id_map<buffer> layer_out(get_chunk_count());
      int j = 0;
      for (int c : layer->chunks) {
        shard_id_t cs((int8_t)c);
        if (!erasures.contains(cs)) {
          if (in.contains(cs))
            layer_in[j] = in.at(cs);
          else
            layer_in[j] = out.at(cs);  // recovered by a previous layer
        } else {
          layer_out[j] = out.at(cs);
        }
        ++j;
      }
      int err = layer->erasure_code->decode_chunks(layer_want, layer_in,
                                                   layer_out);
      if (err) return err;
      for (int c : layer->chunks) erasures.erase(c);
      want_to_read_erasures = shard_id_set();
      for (auto s : want_to_read)
        if (erasures.contains(s)) want_to_read_erasures.insert(s);
      if (want_to_read_erasures.empty()) break;
    }
    return want_to_read_erasures.empty() ? 0 : -EIO;
  }

  // locality-aware source selection (ErasureCodeLrc.cc:578-700, case 1+2)
  int _minimum_to_decode(const shard_id_set &want_to_read,
                         const shard_id_set &available,
                         shard_id_set *minimum) override {
    shard_id_set erasures_want, erasures_not_recovered;
    for (int i = 0; i < (int)get_chunk_count(); i++) {
      if (!available.contains(i)) {
        erasures_not_recovered.insert(i);
        if (want_to_read.contains(i)) erasures_want.insert(i);
      }
    }
    if (erasures_want.empty()) {
      *minimum = want_to_read;
      return 0;
    }
    for (auto layer = layers.rbegin(); layer != layers.rend(); ++layer) {
      shard_id_set layer_want;
      for (auto s : want_to_read)
        if (layer->chunk_set.contains(s)) layer_want.insert(s);
      if (layer_want.empty()) continue;
      bool want_missing = false;
      for (auto s : layer_want)
        if (erasures_want.contains(s)) want_missing = true;
      shard_id_set layer_minimum;
      if (!want_missing) {
        layer_minimum = layer_want;
      } else {
        shard_id_set erasures;
        for (auto s : layer->chunk_set)
          if (erasures_not_recovered.contains(s)) erasures.insert(s);
        if (erasures.size() >
            layer->erasure_code->get_coding_chunk_count())
          continue;  // too many for this layer; hope for an upper layer
        for (auto s : layer->chunk_set)
          if (!erasures_not_recovered.contains(s)) layer_minimum.insert(s);
        for (auto s : erasures) {
          erasures_not_recovered.erase(s);
          erasures_want.erase(s);
        }
      }
      for (auto s : layer_minimum) minimum->insert(s);
    }
    if (erasures_want.empty()) {
      for (auto s : want_to_read) minimum->insert(s);
      for (int i = 0; i < (int)get_chunk_count(); i++)
        if (!available.contains(i)) minimum->erase(i);
      return 0;
    }
    return -EIO;
  }
};

class ErasureCodePluginLrc final : public ErasureCodePlugin {
 public:
  int factory(const std::string &directory, ErasureCodeProfile &profile,
              ErasureCodeInterfaceRef *erasure_code,
              std::ostream *ss) override {
    auto interface = std::make_shared<ErasureCodeLrc>(directory);
    int r = interface->init(profile, ss);
    if (r) return r;
    *erasure_code = interface;
    return 0;
  }
};

}  // namespace

extern "C" {
const char *__erasure_code_version() { return ECX_HARNESS_VERSION; }

int __erasure_code_init(const char *plugin_name, const char *) {
  auto &instance = ErasureCodePluginRegistry::instance();
  auto plugin = std::make_unique<ErasureCodePluginLrc>();
  int r = instance.add(plugin_name, plugin.get());
  if (r == 0) plugin.release();
  return r;
}
}
