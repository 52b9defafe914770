"""Convergence-to-quality run: train a copy task through
the REAL marian-train CLI (text corpus, SentencePiece-free word vocab,
ragged batches, scheduler, validators) until `--valid-metrics bleu-detok`
on a HELD-OUT set reaches the pre-registered target, under
--early-stopping. Pre-registered target: BLEU >= 50 (copy task learns to
~100; 50 is the pass bar).

Usage: python convergence_run.py [--workdir DIR] [--gpu]
Writes the training log to <workdir>/train.log and prints PASS/FAIL.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workdir", default="/tmp/convergence")
    ap.add_argument("--target", type=float, default=50.0)
    ap.add_argument("--after-batches", type=int, default=1500)
    ap.add_argument("--task", choices=["copy", "reverse"], default="copy",
                    help="reverse: target = reversed source (needs real "
                         "attention structure, not identity)")
    args = ap.parse_args()
    os.makedirs(args.workdir, exist_ok=True)
    d = os.path.abspath(args.workdir)

    import torch
    from marian_amd.data.synthetic import write_synthetic_corpus

    # corpus: copy task — train on (src, src); hold out 150 lines
    full_src = os.path.join(d, "full.src")
    write_synthetic_corpus(full_src, os.path.join(d, "full.trg"),
                           6150, 150, 3, 10, seed=99)
    with open(full_src) as f:
        lines = f.read().splitlines()
    if args.task == "reverse":
        trg_lines = [" ".join(reversed(l.split())) for l in lines]
    else:
        trg_lines = lines
    for name, data in (("train.src", lines[:6000]),
                       ("train.trg", trg_lines[:6000]),
                       ("valid.src", lines[6000:]),
                       ("valid.trg", trg_lines[6000:])):
        with open(os.path.join(d, name), "w") as f:
            f.write("\n".join(data) + "\n")

    from marian_amd.training.train import main_train as train_main
    on_gpu = torch.cuda.is_available()
    argv = [
        "--train-sets", os.path.join(d, "train.src"),
        os.path.join(d, "train.trg"),
        "--vocabs", os.path.join(d, "v.yml"), os.path.join(d, "v.yml"),
        "--model", os.path.join(d, "model.npz"),
        "--dim-emb", "256", "--enc-depth", "2", "--dec-depth", "2",
        "--transformer-heads", "4", "--transformer-dim-ffn", "1024",
        "--tied-embeddings-all", "--transformer-dropout", "0.0",
        "--mini-batch", "128" if on_gpu else "32",
        "--maxi-batch", "32",
        "--learn-rate", "0.0005", "--lr-warmup", "40",
        "--label-smoothing", "0.1", "--cost-type", "ce-mean-words",
        "--precision", "bfloat16" if on_gpu else "float32", "float32",
        "--after-batches", str(args.after_batches),
        "--disp-freq", "50u", "--save-freq", "100000u",
        "--valid-sets", os.path.join(d, "valid.src"),
        os.path.join(d, "valid.trg"),
        "--valid-freq", "100u", "--valid-metrics", "bleu-detok",
        "--valid-mini-batch", "32", "--beam-size", "4",
        "--early-stopping", "4",
        "--valid-log", os.path.join(d, "valid.log"),
        "--log", os.path.join(d, "train.log"),
        "--seed", "7", "--overwrite",
        "--cpu-threads", "0" if on_gpu else "1",
    ]
    train_main(argv)

    # read the best BLEU from the validation log
    best = -1.0
    with open(os.path.join(d, "valid.log")) as f:
        for ln in f:
            if "bleu-detok" in ln:
                try:
                    val = float(ln.split("bleu-detok")[1].split(":")[1]
                                .split()[0])
                    best = max(best, val)
                except (IndexError, ValueError):
                    pass
    with open(os.path.join(d, "valid.log")) as f:
        for ln in f.read().splitlines()[-8:]:
            print("VALID:", ln)
    print(f"CONVERGENCE: best bleu-detok = {best:.2f} "
          f"(pre-registered target {args.target}) -> "
          f"{'PASS' if best >= args.target else 'FAIL'}")
    return 0 if best >= args.target else 1


if __name__ == "__main__":
    sys.exit(main())
