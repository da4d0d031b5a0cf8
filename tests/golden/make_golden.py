#!/usr/bin/env python3
"""Generate/validate the golden fixtures under tests/golden/.

Run in the build container where /root/reference is mounted (the GPU box does
NOT have /root/reference — tests must only read the committed fixtures).

Fixtures:
  - wordcount.in            copy of the reference test fixture
                            /root/reference/tests/inputs/wordcount.in
                            (input DATA of the one true in-repo KAT,
                            tests/examples/word_count_test.cpp:36-79)
  - bacon_ipsum_correct.json  the 71-entry word->count table restated from
                            tests/examples/word_count_test.cpp:38-59.
                            This script independently recomputes the counts
                            from wordcount.in with the reference's tokenizer
                            semantics (ReadLines -> tlx::split_view(' '),
                            empty tokens dropped: word_count.hpp:37-45) and
                            asserts the transcribed table matches — i.e. the
                            fixture is pinned BOTH by transcription and by
                            recomputation.
"""
import collections
import json
import os
import shutil
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
REF = "/root/reference"

# Transcribed from /root/reference/tests/examples/word_count_test.cpp:38-59
BACON_TABLE = {
    "alcatra": 32, "amet": 4, "andouille": 16, "bacon": 36,
    "ball": 16, "beef": 40, "belly": 24, "biltong": 24,
    "boudin": 12, "bresaola": 12, "brisket": 24,
    "capicola": 24, "chicken": 4, "chop": 20, "chuck": 24,
    "corned": 16, "cow": 8, "cupim": 20, "dolor": 4,
    "doner": 32, "drumstick": 20, "fatback": 28,
    "filet": 12, "flank": 28, "frankfurter": 12,
    "ground": 8, "ham": 40, "hamburger": 16, "hock": 8,
    "ipsum": 4, "jerky": 28, "jowl": 28, "kevin": 36,
    "kielbasa": 20, "landjaeger": 32, "leberkas": 24,
    "loin": 12, "meatball": 12, "meatloaf": 28,
    "mignon": 12, "pancetta": 24, "pastrami": 16,
    "picanha": 24, "pig": 20, "porchetta": 28, "pork": 64,
    "prosciutto": 24, "ribeye": 20, "ribs": 32, "round": 8,
    "rump": 40, "salami": 20, "sausage": 16, "shank": 12,
    "shankle": 4, "short": 16, "shoulder": 12, "sirloin": 8,
    "spare": 8, "steak": 8, "strip": 8, "swine": 16,
    "t-bone": 16, "tail": 28, "tenderloin": 20, "tip": 16,
    "tongue": 12, "tri-tip": 28, "turducken": 16,
    "turkey": 20, "venison": 20,
}


def tokenize(path):
    """ReadLines -> split_view(' ') semantics: split each line on single
    spaces, drop empty tokens (examples/word_count/word_count.hpp:37-45)."""
    counts = collections.Counter()
    with open(path, "r") as f:
        for line in f:
            for w in line.rstrip("\n").split(" "):
                if w:
                    counts[w] += 1
    return counts


def main():
    src = os.path.join(REF, "tests/inputs/wordcount.in")
    dst = os.path.join(HERE, "wordcount.in")
    if os.path.exists(src):
        shutil.copyfile(src, dst)
        print(f"copied {src} -> {dst}")
    elif not os.path.exists(dst):
        sys.exit("no reference and no committed fixture — cannot proceed")

    counts = tokenize(dst)
    assert dict(counts) == BACON_TABLE, (
        "transcribed table does not match recomputed counts: "
        f"{set(counts.items()) ^ set(BACON_TABLE.items())}")
    assert len(BACON_TABLE) == 71

    out = os.path.join(HERE, "bacon_ipsum_correct.json")
    with open(out, "w") as f:
        json.dump(dict(sorted(BACON_TABLE.items())), f, indent=1)
    print(f"wrote {out} ({len(BACON_TABLE)} entries) — validated against "
          "recomputed counts")


if __name__ == "__main__":
    main()
