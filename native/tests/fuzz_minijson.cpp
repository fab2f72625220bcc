// Deterministic fuzzer for minijson.hpp (the
// hand-rolled JSON parser guards every control link — ckd/ckrt/ckgw —
// and had no sanitizer/fuzz coverage).
//
// No libFuzzer dependency: a seeded xorshift PRNG drives three input
// classes for N iterations under ASAN/UBSAN (make test-asan):
//   1. pure random bytes
//   2. structurally-mutated valid documents (truncation, byte flips,
//      bracket/quote injection, depth bombs)
//   3. round-trip checks: parse(dump(parse(x))) must be stable for
//      valid inputs
// The parser's contract: NEVER crash/overflow — malformed input either
// throws mj::parse_error or yields a value; both are fine.
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "../common/minijson.hpp"

namespace {

uint64_t g_state = 0x9E3779B97F4A7C15ull;

uint64_t rnd() {
  g_state ^= g_state << 13;
  g_state ^= g_state >> 7;
  g_state ^= g_state << 17;
  return g_state;
}

const char* kCorpus[] = {
    R"({"t":"exec","id":"x1","stages":[{"argv":["sh","-c","id"],"uid":0}]})",
    R"({"name":"sb","rootfs":{"lowerdirs":["/"],"upper":"/u"},"netns":true})",
    R"({"a":[1,2.5,-3e10,true,false,null,"é\n\t\"x\""],"b":{"c":{}}})",
    R"([])", R"({})", R"(null)", R"(-0.0)", R"("")", R"(123456789012345678)",
    R"({"deep":[[[[[[[[[[[[[[[[1]]]]]]]]]]]]]]]]})",
    R"({"dup":1,"dup":2})",
};

std::string mutate(const std::string& base) {
  std::string s = base;
  switch (rnd() % 6) {
    case 0:   // truncate
      if (!s.empty()) s.resize(rnd() % s.size());
      break;
    case 1:   // flip bytes
      for (int i = 0; i < 4 && !s.empty(); i++)
        s[rnd() % s.size()] = (char)(rnd() & 0xFF);
      break;
    case 2:   // inject structural chars
      for (int i = 0; i < 3; i++) {
        const char* inj = "{}[]\",:\\";
        s.insert(rnd() % (s.size() + 1), 1, inj[rnd() % 8]);
      }
      break;
    case 3:   // duplicate a slice
      if (s.size() > 2) {
        size_t a = rnd() % s.size(), n = rnd() % (s.size() - a);
        s.insert(rnd() % (s.size() + 1), s.substr(a, n));
      }
      break;
    case 4: {  // depth bomb wrapper
      int depth = 1 + (int)(rnd() % 200);
      std::string pre, post;
      for (int i = 0; i < depth; i++) { pre += "[{\"k\":"; post += "}]"; }
      s = pre + s + post;
      break;
    }
    case 5:   // random garbage
      s.clear();
      for (int i = 0, n = (int)(rnd() % 256); i < n; i++)
        s += (char)(rnd() & 0xFF);
      break;
  }
  return s;
}

}  // namespace

int main(int argc, char** argv) {
  long iters = argc > 1 ? atol(argv[1]) : 100000;
  if (argc > 2) g_state = strtoull(argv[2], nullptr, 10);
  long parsed = 0, rejected = 0;
  for (long i = 0; i < iters; i++) {
    std::string input = mutate(kCorpus[rnd() % (sizeof kCorpus / sizeof *kCorpus)]);
    try {
      mj::Value v = mj::parse(input);
      parsed++;
      // round-trip: dump must be re-parseable and stable
      std::string d1 = v.dump();
      mj::Value v2 = mj::parse(d1);
      std::string d2 = v2.dump();
      if (d1 != d2) {
        fprintf(stderr, "ROUNDTRIP MISMATCH at iter %ld:\n  in: %.200s\n"
                "  d1: %.200s\n  d2: %.200s\n", i, input.c_str(), d1.c_str(),
                d2.c_str());
        return 1;
      }
    } catch (const std::exception&) {
      rejected++;   // clean rejection is a pass
    }
  }
  printf("fuzz_minijson OK: %ld iters (%ld parsed, %ld rejected)\n",
         iters, parsed, rejected);
  return 0;
}
