/* dia_test.cpp — C++ surface tests in the shape of the reference's own
 * tests (tests/api/sort_node_test.cpp, tests/examples/word_count_test.cpp),
 * executing the DOps on the GPU through t9::api. Needs a GPU.
 *
 * Build: hipcc dia_test.cpp -I../../include -L../../thrill_amd -lt9
 *   (done by __graft_entry__.build(); run by tests/test_gpu_dia.py)
 */
#include <t9/dia.hpp>

#include <algorithm>
#include <cstdio>
#include <fstream>
#include <iterator>
#include <map>
#include <random>
#include <sstream>
#include <string>
#include <vector>

static int failures = 0;
#define CHECK(cond)                                                       \
    do {                                                                  \
        if (!(cond)) {                                                    \
            ++failures;                                                   \
            std::fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__,  \
                         #cond);                                          \
        }                                                                 \
    } while (0)

using namespace t9;

/* TeraSort record, as examples/terasort/terasort.cpp:31-43 */
struct Record {
    uint8_t key[10];
    uint8_t value[90];
    bool operator<(const Record& b) const {
        return std::memcmp(this, &b, sizeof(Record)) < 0;  /* total order */
    }
    bool operator==(const Record& b) const {
        return std::memcmp(this, &b, sizeof(Record)) == 0;
    }
} __attribute__((packed));
static_assert(sizeof(Record) == 100, "Record packing");

/* sort_node_test.cpp:25-53: known integers reversed -> exact identity */
static void test_sort_known_integers(api::Context& ctx) {
    const size_t n = 1 << 20;
    auto dia = api::Generate(
        ctx, n, [&](size_t i) { return (uint64_t)(n - 1 - i); });
    auto sorted = dia.Sort(std::less<uint64_t>());
    CHECK(sorted.Size() == n);
    auto out = sorted.AllGather();
    bool ok = true;
    for (size_t i = 0; i < n; ++i) ok &= out[i] == i;
    CHECK(ok);
}

/* sort_node_test.cpp:162-276: degenerate distributions */
static void test_sort_degenerate(api::Context& ctx) {
    for (size_t n : { size_t(0), size_t(1), size_t(3), size_t(1000) }) {
        auto dia = api::Generate(
            ctx, n, [](size_t) { return (uint64_t)42; });
        auto out = dia.Sort().AllGather();
        CHECK(out.size() == n);
        for (auto v : out) CHECK(v == 42);
    }
}

/* record sort against an in-test std::sort oracle (total order) */
static void test_sort_records(api::Context& ctx) {
    const size_t n = 50000;
    std::mt19937_64 rng(0x7421);
    auto gen = [&rng](size_t) {
        Record r;
        uint8_t* p = (uint8_t*)&r;
        for (size_t j = 0; j < sizeof(Record); ++j)
            p[j] = (uint8_t)rng();
        return r;
    };
    std::vector<Record> input;
    input.reserve(n);
    for (size_t i = 0; i < n; ++i) input.push_back(gen(i));

    auto dia = api::FromVector(ctx, input);
    auto out = dia.Sort(api::LexicographicLess<Record>()).AllGather();
    std::vector<Record> expect = input;
    std::sort(expect.begin(), expect.end());
    CHECK(out.size() == expect.size());
    bool ok = true;
    for (size_t i = 0; i < n; ++i) ok &= out[i] == expect[i];
    CHECK(ok);
}

/* word_count_test.cpp:36-79/85-127 pattern: tokenize the bacon-ipsum
 * fixture, reduce (hash, 1) pairs on the GPU, compare with a std::map
 * oracle computed in-test. */
static void test_word_count(api::Context& ctx) {
    std::ifstream in("tests/golden/wordcount.in");
    CHECK(in.good());
    std::vector<std::string> words;
    std::string line;
    while (std::getline(in, line)) {
        std::stringstream ss(line);
        std::string w;
        while (std::getline(ss, w, ' '))
            if (!w.empty()) words.push_back(w);
    }
    CHECK(words.size() > 1000);
    std::map<std::string, size_t> expect;
    for (auto& w : words) expect[w] += 1;
    CHECK(expect.size() == 71);   /* the 71-entry KAT table */

    std::hash<std::string> h;
    std::map<uint64_t, std::string> back;
    std::vector<api::KeyValue> pairs;
    for (auto& w : words) {
        uint64_t k = h(w);
        back[k] = w;
        pairs.push_back(api::KeyValue{ k, 1 });
    }
    CHECK(back.size() == expect.size());   /* hash collision-free here */

    auto reduced = api::ReducePair(api::FromVector(ctx, pairs));
    auto out = reduced.AllGather();
    CHECK(out.size() == expect.size());
    std::map<std::string, size_t> got;
    for (auto& kv : out) got[back[kv.key]] = kv.value;
    CHECK(got == expect);
}

/* the generic string ReduceByKey surface (reference
 * api/reduce_by_key.hpp:241-463 restricted to (string, u64) pairs):
 * additive reduce_fn -> GPU 128-bit path; non-additive reduce_fn ->
 * detected by the algebraic probe, host group-fold with the user's
 * functor. Both must match a std::map oracle. */
static void test_reduce_by_key_string(api::Context& ctx) {
    using P = std::pair<std::string, size_t>;
    std::vector<P> items;
    std::map<std::string, size_t> sums, maxs;
    for (int i = 0; i < 5000; ++i) {
        std::string w = "w" + std::to_string(i % 97);
        size_t v = (size_t)(i % 13) + 1;
        items.push_back(P{ w, v });
        sums[w] += v;
        maxs[w] = std::max(maxs[w], v);
    }
    auto dia = api::FromVector(ctx, items);
    auto keyx = [](const P& p) -> std::string { return p.first; };

    auto add = dia.ReduceByKey(keyx, [](const P& a, const P& b) -> P {
        return P(a.first, a.second + b.second);
    });
    std::map<std::string, size_t> got_add;
    for (auto& p : add.AllGather()) got_add[p.first] = p.second;
    CHECK(got_add == sums);

    auto mx = dia.ReduceByKey(keyx, [](const P& a, const P& b) -> P {
        return P(a.first, std::max(a.second, b.second));
    });
    std::map<std::string, size_t> got_max;
    for (auto& p : mx.AllGather()) got_max[p.first] = p.second;
    CHECK(got_max == maxs);
}

/* WriteBinary/ReadBinary round trip (terasort.cpp:184-200 file mode;
 * on-disk bytes = packed records, data/serialization.hpp:35-48) */
static void test_binary_io(api::Context& ctx) {
    const size_t n = 10000;
    std::mt19937_64 rng(5);
    std::vector<Record> input(n);
    for (auto& r : input)
        for (size_t j = 0; j < sizeof(Record); ++j)
            ((uint8_t*)&r)[j] = (uint8_t)rng();
    auto dia = api::FromVector(ctx, input);
    dia.Sort(api::LexicographicLess<Record>())
        .WriteBinary("/tmp/t9_dia_io-");
    auto back = api::ReadBinary<Record>(
        ctx, { "/tmp/t9_dia_io-0000000000" });
    CHECK(back.Size() == n);
    auto out = back.AllGather();
    std::vector<Record> expect = input;
    std::sort(expect.begin(), expect.end());
    bool ok = true;
    for (size_t i = 0; i < n; ++i) ok &= out[i] == expect[i];
    CHECK(ok);
    std::remove("/tmp/t9_dia_io-0000000000");
}

/* Merge (api/merge.hpp): u64 and whole-record byte-lex variants, A-wins
 * ties (source order among equals) */
static void test_merge(api::Context& ctx) {
    std::vector<uint64_t> a, b;
    for (uint64_t i = 0; i < 20000; ++i) a.push_back(2 * i);
    for (uint64_t i = 0; i < 15000; ++i) b.push_back(3 * i);
    auto m = api::Merge(api::FromVector(ctx, a), api::FromVector(ctx, b));
    auto out = m.AllGather();
    std::vector<uint64_t> expect = a;
    expect.insert(expect.end(), b.begin(), b.end());
    std::sort(expect.begin(), expect.end());
    CHECK(out == expect);

    std::mt19937_64 rng(11);
    std::vector<Record> ra(5000), rb(7000);
    for (auto* v : { &ra, &rb })
        for (auto& r : *v)
            for (size_t j = 0; j < sizeof(Record); ++j)
                ((uint8_t*)&r)[j] = (uint8_t)(rng() & 3);  /* many ties */
    std::sort(ra.begin(), ra.end());
    std::sort(rb.begin(), rb.end());
    auto rm = api::Merge(api::FromVector(ctx, ra),
                         api::FromVector(ctx, rb));
    auto rout = rm.AllGather();
    std::vector<Record> rexpect;
    std::merge(ra.begin(), ra.end(), rb.begin(), rb.end(),
               std::back_inserter(rexpect));   /* std::merge: a-wins */
    CHECK(rout.size() == rexpect.size());
    bool ok = true;
    for (size_t i = 0; i < rout.size(); ++i) ok &= rout[i] == rexpect[i];
    CHECK(ok);
}

/* GroupByKey (api/group_by_key.hpp): per-key value collections */
static void test_group_by_key(api::Context& ctx) {
    const size_t n = 50000;
    std::vector<api::KeyValue> pairs(n);
    for (size_t i = 0; i < n; ++i)
        pairs[i] = api::KeyValue{ i % 97, i };
    auto sums = api::GroupByKey<std::pair<uint64_t, uint64_t> >(
        api::FromVector(ctx, pairs),
        [](uint64_t key, const uint64_t* b, const uint64_t* e) {
            uint64_t s = 0;
            for (const uint64_t* p = b; p != e; ++p) s += *p;
            return std::make_pair(key, s);
        });
    CHECK(sums.size() == 97);
    bool ok = true;
    for (auto& kv : sums) {
        uint64_t expect = 0;
        for (size_t i = kv.first; i < n; i += 97) expect += i;
        ok &= kv.second == expect;
    }
    CHECK(ok);
}

int main() {
    return api::Run([](api::Context& ctx) {
        test_sort_known_integers(ctx);
        test_sort_degenerate(ctx);
        test_sort_records(ctx);
        test_word_count(ctx);
        test_reduce_by_key_string(ctx);
        test_binary_io(ctx);
        test_merge(ctx);
        test_group_by_key(ctx);
        if (failures == 0)
            std::printf("dia_test: all checks passed\n");
        else
            std::printf("dia_test: %d FAILURES\n", failures);
        if (failures) std::exit(1);
    });
}
