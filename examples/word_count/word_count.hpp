/*******************************************************************************
 * examples/word_count/word_count.hpp — port of the reference WordCount
 * pipeline (thrill/examples/word_count/word_count.hpp:25-57) against the
 * t9::api surface. The operator chain — FlatMap tokenizer emitting
 * (word, 1) pairs, then ReduceByKey with the word as key and counter
 * addition as the reduction — is kept statement for statement; what
 * changed and why:
 *   - tlx::split_view is replaced by a plain find-loop (tlx is the
 *     reference's vendored dependency, not part of this framework);
 *   - the InputStack template parameter disappears: the reference's DIA
 *     carries a compile-time lambda-fusion stack (api/dia.hpp:358-405);
 *     this surface fuses host-side Maps eagerly, so DIA<std::string> is
 *     the input type.
 * The ReduceByKey lambdas are verbatim; the GPU path dictionary-encodes
 * the words and reduces on the 128-bit composite table (dia.hpp).
 ******************************************************************************/
#pragma once

#include <t9/dia.hpp>

#include <string>
#include <utility>

namespace examples {
namespace word_count {

using namespace t9;                 // NOLINT (reference: namespace thrill)

using WordCountPair = std::pair<std::string, size_t>;

//! The most basic WordCount user program: reads a DIA containing
//! std::string words, and returns a DIA containing WordCountPairs.
inline auto WordCount(const api::DIA<std::string>& input) {

    auto word_pairs = input.template FlatMap<WordCountPair>(
        [](const std::string& line, auto emit) -> void {
            /* map lambda: emit each word */
            size_t b = 0;
            while (b <= line.size()) {
                size_t e = line.find(' ', b);
                if (e == std::string::npos) e = line.size();
                if (e > b)
                    emit(WordCountPair(line.substr(b, e - b), 1));
                b = e + 1;
            }
        });

    return word_pairs.ReduceByKey(
        [](const WordCountPair& in) -> std::string {
            /* reduction key: the word string */
            return in.first;
        },
        [](const WordCountPair& a, const WordCountPair& b) -> WordCountPair {
            /* associative reduction operator: add counters */
            return WordCountPair(a.first, a.second + b.second);
        });
}

} // namespace word_count
} // namespace examples
