/*******************************************************************************
 * examples/word_count/word_count.cpp — driver for the ported WordCount
 * pipeline (reference: thrill/examples/word_count/word_count.cpp run
 * mode: ReadLines -> WordCount -> print). Reads a text file into a
 * DIA<std::string> of lines, runs the verbatim pipeline from
 * word_count.hpp, and prints "word: count" sorted by word (the
 * reference's word_count_test.cpp compares after sorting, :74).
 ******************************************************************************/

#include "word_count.hpp"

#include <algorithm>
#include <fstream>
#include <iostream>
#include <string>
#include <vector>

using namespace t9;  // NOLINT

int main(int argc, char* argv[]) {
    if (argc < 2) {
        std::cerr << "usage: word_count <input.txt> [output.txt]"
                  << std::endl;
        return 1;
    }
    std::string in_path = argv[1];
    std::string out_path = argc > 2 ? argv[2] : "";

    return api::Run([&](api::Context& ctx) {
        std::vector<std::string> lines;
        {
            std::ifstream f(in_path);
            if (!f) {
                std::cerr << "cannot open " << in_path << std::endl;
                std::exit(1);
            }
            std::string line;
            while (std::getline(f, line)) lines.push_back(line);
        }
        auto input = api::FromVector(ctx, lines);

        auto counts = examples::word_count::WordCount(input).AllGather();

        std::sort(counts.begin(), counts.end());
        std::ostream* os = &std::cout;
        std::ofstream fo;
        if (!out_path.empty()) {
            fo.open(out_path);
            os = &fo;
        }
        for (auto& wc : counts)
            *os << wc.first << ": " << wc.second << std::endl;
    });
}
