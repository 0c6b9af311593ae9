#!/usr/bin/env python3
"""Train the gateway's default token-estimation BPE and export it in
tiktoken file format (base64(token_bytes) + " " + rank per line).

The reference estimates usage with tiktoken cl100k_base when a backend
omits the usage field (llmlb/src/token/mod.rs:217-223).  This
environment has no egress, so the real cl100k table cannot be fetched;
instead we train a byte-level BPE of the same file format on in-image
English+code text.  Deployments can point LLMLB_TIKTOKEN_FILE at a real
cl100k_base.tiktoken for exact parity — the gateway encoder consumes
the standard format.

Run from the repo root:  python gateway/tools/train_bpe.py
Outputs: gateway/data/llmlb.tiktoken, gateway/data/bpe_golden.json
"""
import base64
import json
import os
import sys

from tokenizers import Tokenizer, models, pre_tokenizers, trainers

VOCAB = 24000
OUT = os.path.join(os.path.dirname(__file__), "..", "data")


def corpus_files():
    roots = ["/usr/lib/python3.10", "/usr/share/doc"]
    exts = (".py", ".txt", ".md", ".rst")
    out = []
    for root in roots:
        for dirpath, _dirs, files in os.walk(root):
            for f in files:
                if f.endswith(exts) or "README" in f or f == "copyright":
                    p = os.path.join(dirpath, f)
                    try:
                        if 0 < os.path.getsize(p) < 2_000_000:
                            out.append(p)
                    except OSError:
                        pass
    return out


def main():
    files = corpus_files()
    print(f"training on {len(files)} files")
    tok = Tokenizer(models.BPE())
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    trainer = trainers.BpeTrainer(
        vocab_size=VOCAB, show_progress=False,
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet())
    tok.train(files, trainer)

    # invert the GPT-2 byte<->unicode table to raw bytes
    bs = list(range(ord("!"), ord("~") + 1)) + \
        list(range(0xA1, 0xAD)) + list(range(0xAE, 0x100))
    cs = bs[:] + [256 + i for i in range(256 - len(bs))]
    u2b = {chr(c): b for b, c in zip(bs + [b for b in range(256)
                                          if b not in bs], cs + [])}
    # rebuild properly: gpt2 mapping
    bs2 = bs[:]
    cs2 = bs2[:]
    n = 0
    for b in range(256):
        if b not in bs2:
            bs2.append(b)
            cs2.append(256 + n)
            n += 1
    u2b = {chr(c): b for b, c in zip(bs2, cs2)}

    vocab = tok.get_vocab()  # token-string -> id
    id2tok = {i: t for t, i in vocab.items()}
    lines = []
    for i in range(len(id2tok)):
        t = id2tok[i]
        raw = bytes(u2b[ch] for ch in t)
        lines.append(base64.b64encode(raw).decode() + " " + str(i))
    os.makedirs(OUT, exist_ok=True)
    path = os.path.join(OUT, "llmlb.tiktoken")
    with open(path, "w") as f:
        f.write("\n".join(lines) + "\n")
    print(f"wrote {path} ({len(lines)} tokens)")

    golden = {}
    samples = [
        "Hello, world!",
        "The quick brown fox jumps over the lazy dog.",
        "def fibonacci(n):\n    return n if n < 2 else "
        "fibonacci(n-1) + fibonacci(n-2)\n",
        "Streaming tokens per second is the headline metric "
        "for LLM serving gateways.",
        "   indented\ttext with   runs of whitespace\n\n",
        "numbers 12345 and 2026-09-14 09:30:00",
        "don't can't we'll they're I'm you've he'd",
    ]
    for s in samples:
        golden[s] = len(tok.encode(s).ids)
    with open(os.path.join(OUT, "bpe_golden.json"), "w") as f:
        json.dump(golden, f, indent=1)
    print("golden:", golden)


if __name__ == "__main__":
    sys.exit(main())
