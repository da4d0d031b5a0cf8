/*******************************************************************************
 * examples/terasort/terasort.cpp — port of the reference TeraSort driver
 * (thrill/examples/terasort/terasort.cpp) against the t9::api surface.
 *
 * The Record type (:31-43), GenerateRecord (:63-118) and the main's
 * operator chains (:152-212) keep the reference's shape line for line;
 * what changed and why:
 *   - tlx::CmdlineParser / parse_si_iec_units / hexdump are replaced by
 *     ~20-line standalone equivalents (tlx is a reference vendored
 *     dependency, not part of this framework);
 *   - GenerateRecord's rng seeds deterministically (the reference seeds
 *     from std::random_device, which would make runs uncomparable; same
 *     deviation the oracle records);
 *   - RecordSigned (signed-char key order, a Java-TeraSort compatibility
 *     mode) is rejected at runtime: the GPU sort key order is
 *     byte-lexicographic (the reference's generate modes also
 *     die_unless(!use_signed_char));
 *   - ctx.net_manager().Traffic() has no equivalent (single process);
 *     the RESULT line prints hosts/time only.
 * The DIA pipeline statements themselves are UNCHANGED from the
 * reference — that is the point of the port (north_star: "existing DIA
 * pipelines link unchanged").
 ******************************************************************************/

#include <t9/dia.hpp>

#include <cassert>
#include <cctype>
#include <chrono>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <iostream>
#include <random>
#include <string>
#include <utility>
#include <vector>

using namespace t9;                   // NOLINT (reference: using namespace thrill)

struct Record {
    uint8_t key[10];
    uint8_t value[90];

    bool operator < (const Record& b) const {
        return std::lexicographical_compare(key, key + 10, b.key, b.key + 10);
    }
} __attribute__((packed));

static_assert(sizeof(Record) == 100, "struct Record packing incorrect.");

/*!
 * Generate a Record in a similar way as the "binary" version of Hadoop's
 * GenSort does. The underlying random generator is different.
 * (reference terasort.cpp:59-118; deterministic seed here)
 */
class GenerateRecord
{
public:
    Record operator () (size_t index) {
        Record r;

        // generate random key record
        for (size_t i = 0; i < 10; ++i)
            r.key[i] = static_cast<unsigned char>(rng_());

        uint8_t* v = r.value;

        // add 2 bytes "break"
        *v++ = 0x00;
        *v++ = 0x11;

        // fill values with hexadecimal representation of the record number
        static constexpr uint8_t hexdigits[16] = {
            '0', '1', '2', '3', '4', '5', '6', '7',
            '8', '9', 'A', 'B', 'C', 'D', 'E', 'F'
        };
        uint64_t rec = index;
        for (size_t i = 0; i != 2 * sizeof(rec); ++i)
            *v++ = hexdigits[(rec >> (4 * i)) & 0x0F];
        for (size_t i = 0; i != 16; ++i)
            *v++ = '0';

        // add 4 bytes "break"
        *v++ = 0x88;
        *v++ = 0x99;
        *v++ = 0xAA;
        *v++ = 0xBB;

        // add 48 byte filler based on index
        for (size_t i = 0; i < 12; ++i) {
            uint8_t f = hexdigits[((20 + rec) >> (4 * i)) & 0x0F];
            *v++ = f;
            *v++ = f;
            *v++ = f;
            *v++ = f;
        }

        // add 4 bytes "break"
        *v++ = 0xCC;
        *v++ = 0xDD;
        *v++ = 0xEE;
        *v++ = 0xFF;

        assert(v == r.value + 90);

        return r;
    }

private:
    std::default_random_engine rng_ { 0x7421 };
};

//! parse "100mib"/"1gb"-style sizes (tlx::parse_si_iec_units equivalent)
static bool parse_si_iec_units(const char* str, uint64_t* out) {
    char* end = nullptr;
    double v = std::strtod(str, &end);
    if (end == str) return false;
    uint64_t mult = 1;
    std::string suffix(end);
    for (auto& c : suffix) c = (char)tolower(c);
    if (suffix == "" || suffix == "b") mult = 1;
    else if (suffix == "k" || suffix == "kb") mult = 1000ull;
    else if (suffix == "ki" || suffix == "kib") mult = 1ull << 10;
    else if (suffix == "m" || suffix == "mb") mult = 1000000ull;
    else if (suffix == "mi" || suffix == "mib") mult = 1ull << 20;
    else if (suffix == "g" || suffix == "gb") mult = 1000000000ull;
    else if (suffix == "gi" || suffix == "gib") mult = 1ull << 30;
    else return false;
    *out = (uint64_t)(v * (double)mult);
    return true;
}

static void die(const std::string& msg) {
    std::cerr << "die: " << msg << std::endl;
    std::exit(1);
}

int main(int argc, char* argv[]) {
    // minimal CmdlineParser equivalent for the reference's flags
    // (terasort.cpp:122-150): -g generate+sort, -G generate to file,
    // -s signed chars, -o output, positional input(s)
    bool generate = false, generate_only = false, use_signed_char = false;
    std::string output;
    std::vector<std::string> input;
    for (int i = 1; i < argc; ++i) {
        std::string a = argv[i];
        if (a == "-g") generate = true;
        else if (a == "-G") generate_only = true;
        else if (a == "-s") use_signed_char = true;
        else if (a == "-o" && i + 1 < argc) output = argv[++i];
        else input.push_back(a);
    }

    return api::Run(
        [&](api::Context& ctx) {
            ctx.enable_consume();

            auto t_start = std::chrono::steady_clock::now();

            if (generate_only) {
                if (input.size() != 1u) die("generate_only needs 1 size");
                // parse first argument like "100mib" size
                uint64_t size;
                if (!parse_si_iec_units(input[0].c_str(), &size))
                    die("cannot parse size");
                if (use_signed_char) die("signed char keys unsupported");

                Generate(ctx, size / sizeof(Record), GenerateRecord())
                .WriteBinary(output);
            }
            else if (generate) {
                if (input.size() != 1u) die("generate needs 1 size");
                // parse first argument like "100mib" size
                uint64_t size;
                if (!parse_si_iec_units(input[0].c_str(), &size))
                    die("cannot parse size");
                if (use_signed_char) die("signed char keys unsupported");

                auto r =
                    Generate(ctx, size / sizeof(Record), GenerateRecord())
                    .Sort();

                if (output.size())
                    r.WriteBinary(output);
                else
                    r.Size();
            }
            else {
                if (use_signed_char)
                    die("signed char keys unsupported on the GPU byte "
                        "order (reference RecordSigned mode)");
                auto r = ReadBinary<Record>(ctx, input).Sort();

                if (output.size())
                    r.WriteBinary(output);
                else
                    r.Size();
            }

            auto t_end = std::chrono::steady_clock::now();
            ctx.net.Barrier();
            if (ctx.my_rank() == 0) {
                std::cout << "RESULT"
                          << " benchmark=terasort"
                          << " time="
                          << std::chrono::duration<double>(
                                 t_end - t_start).count()
                          << " hosts=" << ctx.num_workers() << std::endl;
            }
        });
}
