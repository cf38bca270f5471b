"""Deterministic offline BPE training (no network, no data files).

Builds the merge table committed at sentio_amd/engines/assets/bpe_merges.json.
Corpus: the repo's own English prose (prompt templates, README) plus a
seeded synthetic query/document mix over a common-English wordlist — enough
to learn ordinary English subwords (~4 chars/token) deterministically.

Re-running this script reproduces the identical asset (seeded RNG, stable
tie-breaking on (count, pair) ordering).

Usage: python scripts/train_tokenizer.py [--merges 4096]
"""

from __future__ import annotations

import argparse
import collections
import json
import os
import re
import sys

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

COMMON = (
    "the of and to in a is that it for on with as are was be this have from "
    "or by not word but what some can out other were all there when up use "
    "your how said each she which do their time if will way about many then "
    "them would write like so these her long make thing see him two has look "
    "more day could go come did my no most people over know water than call "
    "first who may down side been now find any new work part take get place "
    "made live where after back little only round man year came show every "
    "good me give our under name very through just form much great think say "
    "help low line before turn cause same mean differ move right boy old too "
    "does tell sentence set three want air well also play small end put home "
    "read hand port large spell add even land here must big high such follow "
    "act why ask men change went light kind off need house picture try us "
    "again animal point mother world near build self earth father head stand "
    "own page should country found answer school grow study still learn plant "
    "cover food sun four between state keep eye never last let thought city "
    "tree cross farm hard start might story saw far sea draw left late run "
    "while press close night real life few north open seem together next "
    "white children begin got walk example ease paper group always music "
    "those both mark often letter until mile river car feet care second book "
    "carry took science eat room friend began idea fish mountain stop once "
    "base hear horse cut sure watch color face wood main enough plain girl "
    "usual young ready above ever red list though feel talk bird soon body "
    "dog family direct pose leave song measure door product black short "
    "numeral class wind question happen complete ship area half rock order "
    "fire south problem piece told knew pass since top whole king space heard "
    "best hour better true during hundred five remember step early hold west "
    "ground interest reach fast verb sing listen six table travel less "
    "morning ten simple several vowel toward war lay against pattern slow "
    "center love person money serve appear road map rain rule govern pull "
    "cold notice voice unit power town fine certain fly fall lead cry dark "
    "machine note wait plan figure star box noun field rest correct able "
    "pound done beauty drive stood contain front teach week final gave green "
    "oh quick develop ocean warm free minute strong special mind behind "
    "clear tail produce fact street inch multiply nothing course stay wheel "
    "full force blue object decide surface deep moon island foot system busy "
    "test record boat common gold possible plane stead dry wonder laugh "
    "thousand ago ran check game shape equate hot miss brought heat snow "
    "tire bring yes distant fill east paint language among"
).split()

TECH = ("retrieval embedding document query index search rank rerank fusion "
        "vector cosine similarity token tokens model generation answer "
        "context chunk passage relevance score dense sparse hybrid cache "
        "server request response latency throughput pipeline verify").split()


def build_corpus() -> str:
    parts: list[str] = []
    for name in ("prompts", "."):
        d = os.path.join(ROOT, name)
        if not os.path.isdir(d):
            continue
        for f in sorted(os.listdir(d)):
            if f.endswith(".md") and f.upper() not in ("SNIPPETS.MD",):
                try:
                    parts.append(open(os.path.join(d, f)).read())
                except OSError:
                    pass
    rng = np.random.RandomState(1234)
    words = COMMON + TECH
    for _ in range(4000):
        n = rng.randint(8, 26)
        parts.append(" ".join(rng.choice(words, size=n)) + ".")
    for i in range(800):
        parts.append(f"what does {rng.choice(words)} mean for "
                     f"{rng.choice(words)} {rng.choice(TECH)}?")
    return "\n".join(parts)


def train(corpus: str, n_merges: int) -> list[list[int]]:
    from sentio_amd.engines.bpe import _PRETOKEN
    from sentio_amd.engines.tokenizer import BYTE_OFFSET

    MERGE_OFFSET = BYTE_OFFSET + 256
    word_counts = collections.Counter(
        m.group(0) for m in _PRETOKEN.finditer(corpus.encode("utf-8")))
    words = [([BYTE_OFFSET + b for b in w], c)
             for w, c in sorted(word_counts.items())]
    merges: list[list[int]] = []
    for rank in range(n_merges):
        pair_counts: collections.Counter = collections.Counter()
        for ids, c in words:
            for i in range(len(ids) - 1):
                pair_counts[(ids[i], ids[i + 1])] += c
        if not pair_counts:
            break
        # stable argmax: highest count, then smallest pair ids
        best = min(pair_counts.items(), key=lambda kv: (-kv[1], kv[0]))[0]
        if pair_counts[best] < 2:
            break
        tid = MERGE_OFFSET + rank
        merges.append([best[0], best[1]])
        for wi, (ids, c) in enumerate(words):
            i = 0
            out = []
            while i < len(ids):
                if (i + 1 < len(ids) and ids[i] == best[0]
                        and ids[i + 1] == best[1]):
                    out.append(tid)
                    i += 2
                else:
                    out.append(ids[i])
                    i += 1
            words[wi] = (out, c)
    return merges


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--merges", type=int, default=4096)
    args = ap.parse_args()
    corpus = build_corpus()
    print(f"corpus: {len(corpus)} chars")
    merges = train(corpus, args.merges)
    out = os.path.join(ROOT, "sentio_amd", "engines", "assets",
                       "bpe_merges.json")
    os.makedirs(os.path.dirname(out), exist_ok=True)
    with open(out, "w") as f:
        json.dump({"merges": merges, "corpus_chars": len(corpus),
                   "trained_by": "scripts/train_tokenizer.py"}, f)
    # report compression on held-out-ish text
    from sentio_amd.engines.bpe import BPETokenizer

    tok = BPETokenizer(merges)
    sample = corpus[: 20000]
    ids = tok.encode(sample, None)
    print(f"merges: {len(merges)}  vocab: {tok.vocab_size}  "
          f"chars/token: {len(sample) / max(len(ids), 1):.2f}")
    assert tok.decode(ids) == sample


if __name__ == "__main__":
    main()
