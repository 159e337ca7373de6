"""Topic-model text preprocessor (parity with the reference
data/proc_text_topic.py: lowercase alpha tokens, stop-word filter,
markup-line skip, frequency-capped vocab, then per-document term-count
rows over that vocab). Rewritten for this framework: argparse CLI,
streaming two-pass, and an extra --sparse mode emitting the
(doc, word, count) triples the PLSA/GMM trainers consume directly
(lightctr_amd.models.plsa.PLSAModel.fit).

Output formats:
  vocab file:    "<id> <term> <corpus count>" per line (reference shape)
  dense mode:    one row per kept document: counts for every vocab id
  sparse mode:   "doc word count" triples (one per line)
"""
import argparse
import sys
from collections import Counter

STOPWORDS = frozenset(
    "a the of to an but or its about would and in that is are be been "
    "will this was for on as from at by with have which has had were it "
    "not".split())


def tokens(line: str):
    for raw in line.rstrip().split(" "):
        t = raw.lower()
        if not t or not t.isalpha() or t in STOPWORDS:
            continue
        yield t


def is_markup(line: str) -> bool:
    return "<" in line and ">" in line


def build_vocab(path: str, vocab_size: int):
    freq = Counter()
    with open(path) as f:
        for line in f:
            if is_markup(line):
                continue
            freq.update(tokens(line))
    kept = freq.most_common(vocab_size)
    term_id = {t: i for i, (t, _) in enumerate(kept)}
    return term_id, dict(kept)


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("input")
    ap.add_argument("vocab_size", type=int)
    ap.add_argument("--vocab-out", default="vocab.txt")
    ap.add_argument("--train-out", default="train_topic.csv")
    ap.add_argument("--sparse", action="store_true",
                    help="emit 'doc word count' triples instead of dense "
                         "count rows")
    args = ap.parse_args(argv)

    term_id, freq = build_vocab(args.input, args.vocab_size)
    with open(args.vocab_out, "w") as out:
        for term, tid in sorted(term_id.items(), key=lambda kv: kv[1]):
            out.write(f"{tid} {term} {freq[term]}\n")
    print(f"vocab: {len(term_id)} terms -> {args.vocab_out}")

    n_docs = 0
    with open(args.input) as f, open(args.train_out, "w") as out:
        for line in f:
            if is_markup(line):
                continue
            tf = Counter(t for t in tokens(line) if t in term_id)
            if not tf:
                continue
            if args.sparse:
                for t, c in sorted(tf.items(), key=lambda kv: term_id[kv[0]]):
                    out.write(f"{n_docs} {term_id[t]} {c}\n")
            else:
                row = [0] * len(term_id)
                for t, c in tf.items():
                    row[term_id[t]] = c
                out.write(" ".join(map(str, row)) + "\n")
            n_docs += 1
    print(f"docs: {n_docs} -> {args.train_out}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
