"""UNK replacement using the attention alignment stream.

Behavioural port of scripts/replace_unk.py:16-48. The generator writes
"word [srcpos]" interleaved output (gen driver, gen.py:88-98); this module
parses that stream and replaces each 'UNK' with the source word at the
attended position, skipping '<EOS>' tokens on both sides.
"""

import re


def replace_unk_line(summary_line, source_words, extractive=False,
                     remove_eos=True):
    toks = summary_line.strip().split()
    y = toks[::2]
    pos = [int(re.sub(r"\[|\]", "", p)) for p in toks[1::2]]
    out = []
    for a, b in zip(y, pos):
        if remove_eos and a == "<EOS>":
            continue
        if not extractive:
            if a == "UNK" and b < len(source_words):
                if source_words[b] == "<EOS>":
                    continue
                out.append(source_words[b])
            else:
                out.append(a)
        else:
            out.append(a)
    return " ".join(out)


def replace_unk_files(corpus_path, summary_path, out_path, extractive=False,
                      remove_eos=True):
    all_words = []
    with open(corpus_path) as f:
        for line in f:
            all_words.append(line.strip().split())
    with open(out_path, "w") as fo, open(summary_path) as f:
        for line, words in zip(f, all_words):
            fo.write(replace_unk_line(line, words, extractive, remove_eos))
            fo.write("\n")
