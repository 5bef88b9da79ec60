// registry_selftest.cc — exercises the plugin registry's dlopen semantics
// and the oracle fixture plugin's interface conformance on CPU. Mirrors the
// reference's registry tests (src/test/erasure-code/TestErasureCodePlugin.cc
// failure modes) and the round-trip style of TestErasureCodeJerasure.cc.
// Exit code 0 = all checks passed; prints one line per check.
#include <cerrno>
#include <cstring>
#include <iostream>
#include <random>
#include <sstream>

#include "erasure_code_plugin.h"

using namespace ecx;

static int failures = 0;

#define CHECK(cond, name)                                       \
  do {                                                          \
    if (cond) {                                                 \
      std::cout << "ok " << name << "\n";                       \
    } else {                                                    \
      std::cout << "FAIL " << name << "\n";                     \
      failures++;                                               \
    }                                                           \
  } while (0)

int main(int argc, char **argv) {
  std::string dir = argc > 1 ? argv[1] : ".";
  auto &reg = ErasureCodePluginRegistry::instance();
  reg.disable_dlclose = true;
  std::stringstream ss;

  // --- failure modes (ErasureCodePlugin.cc:138-206 semantics) ---
  {
    ErasureCodeProfile p{{"k", "2"}, {"m", "1"}};
    ErasureCodeInterfaceRef ec;
    int r = reg.factory("does_not_exist", dir, p, &ec, &ss);
    CHECK(r == -EIO, "dlopen-missing-file => -EIO");
  }
  {
    ErasureCodeProfile p{{"k", "2"}, {"m", "1"}};
    ErasureCodeInterfaceRef ec;
    int r = reg.factory("fix_missing_version", dir, p, &ec, &ss);
    CHECK(r == -EXDEV, "missing __erasure_code_version => -EXDEV");
    r = reg.factory("fix_bad_version", dir, p, &ec, &ss);
    CHECK(r == -EXDEV, "wrong version string => -EXDEV");
    r = reg.factory("fix_missing_init", dir, p, &ec, &ss);
    CHECK(r == -ENOENT, "missing __erasure_code_init => -ENOENT");
    r = reg.factory("fix_fail_init", dir, p, &ec, &ss);
    CHECK(r == -ESRCH, "init failure propagates");
    r = reg.factory("fix_no_register", dir, p, &ec, &ss);
    CHECK(r == -EBADF, "init that does not register => -EBADF");
  }

  // --- oracle fixture plugin: load, profile echo, round trip ---
  ErasureCodeInterfaceRef ec;
  {
    ErasureCodeProfile p{{"k", "4"}, {"m", "2"},
                         {"technique", "reed_sol_van"}};
    int r = reg.factory("oracle", dir, p, &ec, &ss);
    CHECK(r == 0, "oracle plugin factory");
    if (r != 0) {
      std::cerr << ss.str() << "\n";
      return 1;
    }
    CHECK(p.at("k") == "4" && p.count("crush-root") == 1,
          "profile defaults echoed back");
    CHECK(ec->get_chunk_count() == 6, "chunk count");
  }
  {
    // encode/decode round trip with every 2-erasure pattern
    const unsigned C = ec->get_chunk_size(4 * 4096);
    buffer in = buffer::create_aligned(4 * 4096);
    std::mt19937_64 rng(0xEC);
    for (size_t i = 0; i < in.length() / 8; i++)
      ((uint64_t *)in.c_str())[i] = rng();
    shard_id_set want;
    for (int i = 0; i < 6; i++) want.insert(i);
    shard_id_map<buffer> encoded(6);
    int r = ec->encode(want, in, &encoded);
    CHECK(r == 0 && encoded.size() == 6 && encoded.at(0).length() == C,
          "encode");
    // systematic prefix: data chunks hold the input verbatim
    CHECK(std::memcmp(encoded.at(0).c_str(), in.c_str(), C) == 0 &&
              std::memcmp(encoded.at(3).c_str(), in.c_str() + 3 * C,
                          in.length() - 3 * C) == 0,
          "systematic prefix");
    bool all_ok = true;
    for (int e1 = 0; e1 < 6; e1++)
      for (int e2 = e1 + 1; e2 < 6; e2++) {
        shard_id_map<buffer> chunks = encoded;
        chunks.erase(e1);
        chunks.erase(e2);
        shard_id_map<buffer> decoded(6);
        shard_id_set want_read;
        want_read.insert(e1);
        want_read.insert(e2);
        if (ec->decode(want_read, chunks, &decoded, 0) != 0) all_ok = false;
        for (int e : {e1, e2})
          if (std::memcmp(decoded.at(e).c_str(), encoded.at(e).c_str(), C))
            all_ok = false;
      }
    CHECK(all_ok, "exhaustive 2-erasure decode round trip");
  }
  {
    // parity-delta conformance: delta path == full re-encode
    ErasureCodeProfile p{{"k", "3"}, {"m", "2"},
                         {"technique", "cauchy"}};
    ErasureCodeInterfaceRef ec2;
    int r = reg.factory("oracle", dir, p, &ec2, &ss);
    CHECK(r == 0, "oracle cauchy factory");
    unsigned C = 4096;
    std::mt19937_64 rng(1);
    shard_id_map<buffer> in(5), out(5);
    for (int i = 0; i < 3; i++) {
      buffer b = buffer::create_aligned(C);
      for (size_t w = 0; w < C / 8; w++) ((uint64_t *)b.c_str())[w] = rng();
      in[i] = b;
    }
    for (int j = 3; j < 5; j++) out[j] = buffer::create_aligned(C);
    ec2->encode_chunks(in, out);
    // mutate chunk 1 via delta
    buffer newc = buffer::create_aligned(C);
    for (size_t w = 0; w < C / 8; w++) ((uint64_t *)newc.c_str())[w] = rng();
    buffer delta = buffer::create_aligned(C);
    ec2->encode_delta(in.at(1), newc, &delta);
    shard_id_map<buffer> din(5), dout(5);
    din[1] = delta;
    dout[3] = out.at(3);
    dout[4] = out.at(4);
    ec2->apply_delta(din, dout);
    in[1] = newc;
    shard_id_map<buffer> out2(5);
    for (int j = 3; j < 5; j++) out2[j] = buffer::create_aligned(C);
    ec2->encode_chunks(in, out2);
    bool same = !std::memcmp(out.at(3).c_str(), out2.at(3).c_str(), C) &&
                !std::memcmp(out.at(4).c_str(), out2.at(4).c_str(), C);
    CHECK(same, "parity delta == re-encode");
  }
  {
    // bitmatrix (cauchy_orig/cauchy_good) parity delta == re-encode:
    // schedule-delta semantics on CPU via the oracle fixture
    // (schedule_apply_delta, ErasureCodeJerasure.cc:348-377)
    for (const char *tech : {"cauchy_orig", "cauchy_good"}) {
      ErasureCodeProfile p{{"k", "4"}, {"m", "3"}, {"technique", tech},
                           {"packetsize", "128"}};
      ErasureCodeInterfaceRef ec3;
      int r = reg.factory("oracle", dir, p, &ec3, &ss);
      CHECK(r == 0, std::string("oracle ") + tech + " factory");
      if (r) continue;
      const unsigned C = 8 * 128 * 2;  // 2 superwords
      std::mt19937_64 rng(7);
      shard_id_map<buffer> in(7), out(7);
      for (int i = 0; i < 4; i++) {
        buffer b = buffer::create_aligned(C);
        for (size_t w = 0; w < C / 8; w++)
          ((uint64_t *)b.c_str())[w] = rng();
        in[i] = b;
      }
      for (int j = 4; j < 7; j++) out[j] = buffer::create_aligned(C);
      CHECK(ec3->encode_chunks(in, out) == 0,
            std::string(tech) + " encode");
      buffer newc = buffer::create_aligned(C);
      for (size_t w = 0; w < C / 8; w++)
        ((uint64_t *)newc.c_str())[w] = rng();
      buffer delta = buffer::create_aligned(C);
      ec3->encode_delta(in.at(2), newc, &delta);
      shard_id_map<buffer> din(7), dout(7);
      din[2] = delta;
      for (int j = 4; j < 7; j++) dout[j] = out.at(j);
      ec3->apply_delta(din, dout);
      in[2] = newc;
      shard_id_map<buffer> out2(7);
      for (int j = 4; j < 7; j++) out2[j] = buffer::create_aligned(C);
      ec3->encode_chunks(in, out2);
      bool same = true;
      for (int j = 4; j < 7; j++)
        same &= !std::memcmp(out.at(j).c_str(), out2.at(j).c_str(), C);
      CHECK(same, std::string(tech) + " schedule delta == re-encode");
    }
  }
  {
    // minimum_to_decode semantics (ErasureCode.cc:154-170)
    shard_id_set want, avail, minimum;
    want.insert(0);
    for (int i : {0, 1, 2, 3, 4, 5}) avail.insert(i);
    ec->minimum_to_decode(want, avail, minimum, nullptr);
    CHECK(minimum.size() == 1 && minimum.contains(0),
          "minimum == want when available");
    shard_id_set avail2, min2;
    for (int i : {1, 2, 4, 5}) avail2.insert(i);
    ec->minimum_to_decode(want, avail2, min2, nullptr);
    CHECK(min2.size() == 4, "minimum == first k available on erasure");
  }

  {
    // LRC decode_chunks from a RAW caller (not ErasureCode::_decode, which
    // invents buffers for all k+m chunks): chunks in neither map must not
    // crash the plugin — layers that lack a backing buffer are skipped and
    // the call returns 0 (recovered) or -EIO, never throws
    ErasureCodeProfile p{{"k", "4"}, {"m", "2"}, {"l", "3"},
                         {"lrc-default-plugin", "oracle"}};
    ErasureCodeInterfaceRef lrc;
    int r = reg.factory("lrc", dir, p, &lrc, &ss);
    CHECK(r == 0, "lrc factory (oracle sub-plugin)");
    if (r == 0) {
      const unsigned n = lrc->get_chunk_count();
      const unsigned K = lrc->get_data_chunk_count();
      const unsigned C = lrc->get_chunk_size(4096 * 4);
      // raw positions: chunk_mapping[logical shard] = raw chunk index;
      // data occupies the mapping's 'D' positions, not 0..K-1
      auto &cm = lrc->get_chunk_mapping();
      std::vector<int> data_pos;
      std::vector<bool> is_data(n, false);
      for (unsigned i = 0; i < K; i++) {
        data_pos.push_back((int)cm[i]);
        is_data[(int)cm[i]] = true;
      }
      std::mt19937 rng(0xECEC);
      shard_id_map<buffer> enc_in(n), enc_out(n);
      for (int dp : data_pos) {
        buffer b = buffer::create_aligned(C);
        for (unsigned x = 0; x < C; x++) b.c_str()[x] = (uint8_t)rng();
        enc_in[dp] = b;
      }
      for (unsigned i = 0; i < n; i++)
        if (!is_data[i]) enc_out[i] = buffer::create_aligned(C);
      CHECK(lrc->encode_chunks(enc_in, enc_out) == 0, "lrc encode for raw test");
      const int d0 = data_pos[0];
      // erase data chunk d0; supply buffers for only two other data
      // chunks — every layer containing d0 also contains unbacked chunks
      shard_id_set want;
      want.insert(d0);
      shard_id_map<buffer> in(n), out(n);
      in[data_pos[1]] = enc_in.at(data_pos[1]);
      in[data_pos[2]] = enc_in.at(data_pos[2]);
      out[d0] = buffer::create_aligned(C);
      bool threw = false;
      int dr = 0;
      try {
        dr = lrc->decode_chunks(want, in, out);
      } catch (...) {
        threw = true;
      }
      CHECK(!threw, "lrc raw decode_chunks with missing buffers: no throw");
      // either the plugin reports failure, or (if some backed layer could
      // recover) the result must be byte-exact — never a crash or garbage
      CHECK(threw || dr != 0 ||
                !std::memcmp(out.at(d0).c_str(), enc_in.at(d0).c_str(), C),
            "lrc raw decode: error or exact recovery, nothing in between");
      // with every other chunk backed, recovery succeeds and is exact
      shard_id_map<buffer> in2(n), out2(n);
      for (unsigned i = 0; i < n; i++) {
        if ((int)i == d0) continue;
        if (enc_in.contains(i)) in2[i] = enc_in.at(i);
        else in2[i] = enc_out.at(i);
      }
      out2[d0] = buffer::create_aligned(C);
      int dr2 = lrc->decode_chunks(want, in2, out2);
      CHECK(dr2 == 0 && !std::memcmp(out2.at(d0).c_str(),
                                     enc_in.at(d0).c_str(), C),
            "lrc raw decode with full buffers recovers the erased chunk");
    }
  }

  {
    // preload: the OSD's osd_erasure_code_plugins startup list
    // (ErasureCodePlugin.cc:208-224 calls load() per csv name). shec and
    // clay are not yet loaded here => fresh preload succeeds; preloading
    // an already-registered name goes through load()->init()->add() and
    // returns -EEXIST, exactly as the reference would (preload is a
    // startup-time call, before any factory)
    std::stringstream pss;
    int r = reg.preload("shec,clay", dir, &pss);
    CHECK(r == 0, "preload csv loads and registers each fresh plugin");
    CHECK(reg.get("shec") != nullptr && reg.get("clay") != nullptr,
          "preloaded plugins visible in the registry");
    r = reg.preload("oracle", dir, &pss);
    CHECK(r == -EEXIST, "re-preloading a registered plugin => -EEXIST");
    r = reg.preload("no_such_plugin", dir, &pss);
    CHECK(r != 0, "preload propagates a load failure");
  }

  std::cout << (failures ? "FAILURES: " : "all ok: ") << failures << "\n";
  return failures ? 1 : 0;
}
