"""Train the shipped BPE tokenizer artifact (offline — no network).

The reference embeds with a pretrained bge-m3 GGUF vocab (llama.cpp);
this environment has no network to fetch it, so we train a real BPE
vocab with the `tokenizers` library on local text (repo sources, docs,
Python stdlib docstrings) and ship the JSON artifact in-tree. The
special-token layout matches XLM-R (<s>=0 <pad>=1 </s>=2 <unk>=3) so
ids drop into the bge-m3 embedding table unchanged; a user with the
real bge-m3 tokenizer.json can point NORNICDB_TOKENIZER at it and get
exact vocab compatibility through the same HFTokenizer code path.

Usage: python scripts/train_tokenizer.py [--vocab 32768]
Writes nornicdb_amd/embed/vocab/nornic_bpe.json
"""
import argparse
import glob
import io
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def corpus_texts():
    pats = ["*.md", "nornicdb_amd/**/*.py", "tests/*.py", "scripts/*.py",
            "docs/**/*.md", "nornicdb_amd/csrc/*.hip",
            "nornicdb_amd/csrc/*.cpp"]
    for p in pats:
        for fn in glob.glob(os.path.join(ROOT, p), recursive=True):
            try:
                yield open(fn, errors="ignore").read()
            except OSError:
                pass
    # python stdlib docstrings: plenty of natural English
    import pydoc
    import pkgutil
    count = 0
    argv_save, sys.argv = sys.argv, [sys.argv[0]]  # imports must not see argv
    skip = {"setup", "antigravity", "this", "idlelib", "turtle", "tkinter",
            "bench", "conftest"}
    for m in sorted(pkgutil.iter_modules(), key=lambda x: x.name):
        if count > 250:
            break
        if m.name in skip or m.name.startswith("_"):
            continue
        try:
            mod = __import__(m.name)
            doc = pydoc.render_doc(mod, renderer=pydoc.plaintext)
            if doc:
                yield doc
                count += 1
        except BaseException:
            continue
    sys.argv = argv_save


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--vocab", type=int, default=32768)
    args = ap.parse_args()
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers, \
        processors, decoders, normalizers

    tok = Tokenizer(models.BPE(unk_token="<unk>"))
    tok.normalizer = normalizers.NFKC()
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=True)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=args.vocab,
        special_tokens=["<s>", "<pad>", "</s>", "<unk>"],
        min_frequency=2,
        show_progress=False,
    )
    texts = list(corpus_texts())
    print(f"training on {sum(len(t) for t in texts) / 1e6:.1f} MB "
          f"from {len(texts)} documents ...")
    tok.train_from_iterator(texts, trainer)
    tok.post_processor = processors.TemplateProcessing(
        single="<s> $A </s>",
        pair="<s> $A </s> </s> $B </s>",
        special_tokens=[("<s>", 0), ("</s>", 2)],
    )
    out = os.path.join(ROOT, "nornicdb_amd", "embed", "vocab")
    os.makedirs(out, exist_ok=True)
    path = os.path.join(out, "nornic_bpe.json")
    tok.save(path)
    print(f"wrote {path} ({os.path.getsize(path) / 1e6:.2f} MB, "
          f"vocab {tok.get_vocab_size()})")
    enc = tok.encode("NornicDB embeds documents with bge-m3 on MI355X.")
    print("sample:", enc.ids[:16])


if __name__ == "__main__":
    main()
