/* io_bench — measure WriteBinary/ReadBinary throughput (SURVEY.md §8f
 * item 2: the reference terasort's real-file mode, api/write_binary.hpp /
 * api/read_binary.hpp / vfs/sys_file.cpp). Generates records ON DEVICE
 * (t9_gen_records), round-trips them through the t9::api file format
 * (packed fixed-size records, bit-compatible with the reference's
 * files), and reports GB/s for the write (fsync included) and the read
 * (after an explicit page-cache drop when running as root). The backing
 * medium is printed so the figure can be judged against it — the GPU
 * boxes mount an overlay on loop devices, not raw NVMe, so the medium
 * line matters.
 *
 * usage: io_bench <bytes> <path-prefix> */

#include <t9/dia.hpp>

#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <fstream>
#include <iostream>
#include <string>
#include <vector>

#include <fcntl.h>
#include <unistd.h>

using namespace t9;  // NOLINT

struct Record {
    uint8_t key[10];
    uint8_t value[90];
} __attribute__((packed));

static double now() {
    return std::chrono::duration<double>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
}

int main(int argc, char* argv[]) {
    uint64_t bytes = argc > 1 ? std::strtoull(argv[1], nullptr, 10)
                              : (1ull << 32);
    std::string prefix = argc > 2 ? argv[2] : "/tmp/t9_io_bench-";
    const uint64_t n = bytes / sizeof(Record);

    return api::Run([&](api::Context& ctx) {
        // generate on device, wrap the buffer as a DIA
        auto hold = std::make_shared<api::DeviceBuf>(n * sizeof(Record));
        T9_DIA_TRY(t9_gen_records(ctx.native(), (uint8_t*)hold->ptr, 0, n,
                                  0x7421, ctx.stream()));
        T9_DIA_HIP(hipStreamSynchronize(ctx.stream()));
        api::DIA<Record> data(&ctx, hold, n);

        double t0 = now();
        data.WriteBinary(prefix);
        // fsync the written file
        {
            char name[512];
            std::snprintf(name, sizeof(name), "%s%010zu", prefix.c_str(),
                          ctx.my_rank());
            int fd = ::open(name, O_RDONLY);
            if (fd >= 0) {
                ::fsync(fd);
                ::close(fd);
            }
        }
        ::sync();
        double t_write = now() - t0;

        // drop the page cache (root) so the read hits the medium
        bool dropped = false;
        {
            std::ofstream f("/proc/sys/vm/drop_caches");
            if (f) {
                f << "3\n";
                dropped = (bool)f;
            }
        }

        char name[512];
        std::snprintf(name, sizeof(name), "%s%010zu", prefix.c_str(),
                      ctx.my_rank());
        t0 = now();
        auto back = api::ReadBinary<Record>(
            ctx, std::vector<std::string>{ name });
        double t_read = now() - t0;

        if (back.Size() != n) {
            std::cerr << "round trip size mismatch" << std::endl;
            std::exit(1);
        }
        // spot-check bytes
        auto h0 = data.AllGather();
        auto h1 = back.AllGather();
        for (uint64_t i = 0; i < n; i += (n > 1000 ? n / 1000 : 1)) {
            if (std::memcmp(&h0[i], &h1[i], sizeof(Record)) != 0) {
                std::cerr << "round trip data mismatch at " << i
                          << std::endl;
                std::exit(1);
            }
        }
        ::unlink(name);

        double gb = (double)(n * sizeof(Record)) / 1e9;
        std::cout << "{\"bytes\": " << n * sizeof(Record)
                  << ", \"write_GBps\": " << gb / t_write
                  << ", \"read_GBps\": " << gb / t_read
                  << ", \"cache_dropped\": " << (dropped ? "true" : "false")
                  << ", \"path\": \"" << prefix << "\"}" << std::endl;
    });
}
