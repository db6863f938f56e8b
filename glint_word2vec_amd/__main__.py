"""CLI — the spark-submit-equivalent entry points of the reference
(README.md:30-57 of the reference: train via spark-submit, PS cluster via
glint.Main) collapse to one command on a single MI355X node:

  python -m glint_word2vec_amd train  CORPUS MODEL_DIR [knobs...]
  python -m glint_word2vec_amd similar MODEL_DIR WORD [-n N]
  python -m glint_word2vec_amd analogy MODEL_DIR POS... -m NEG... [-n N]
  python -m glint_word2vec_amd export  MODEL_DIR VECTORS_TXT
  python -m glint_word2vec_amd info    MODEL_DIR

Multi-GPU training: launch the same train command under torchrun
(one rank per GPU, RCCL):
  torchrun --standalone --nproc-per-node 8 -m glint_word2vec_amd train ...
"""
from __future__ import annotations

import argparse
import logging
import sys

from . import GlintWord2Vec, GlintWord2VecModel


def _add_train(sub):
    p = sub.add_parser("train", help="fit a model on a text corpus")
    p.add_argument("corpus", help="text file: one sentence per line")
    p.add_argument("model", help="output model directory")
    p.add_argument("--vector-size", type=int, default=100)
    p.add_argument("--learning-rate", type=float, default=0.01875)
    p.add_argument("--iterations", type=int, default=1)
    p.add_argument("--window", type=int, default=5)
    p.add_argument("--negatives", type=int, default=5)
    p.add_argument("--min-count", type=int, default=5)
    p.add_argument("--subsample", type=float, default=1e-6)
    p.add_argument("--seed", type=int, default=None)
    p.add_argument("--unigram-table-size", type=int, default=100_000_000)
    p.add_argument("--max-sentence-length", type=int, default=1000)
    p.add_argument("--dtype", choices=["auto", "float32", "bfloat16"],
                   default="auto")
    p.add_argument("--device", choices=["auto", "cpu", "cuda"], default="auto")
    p.add_argument("--engine",
                   choices=["auto", "fused", "dim", "row", "dp"],
                   default="auto")
    p.add_argument("--window-mode", choices=["canonical", "reference"],
                   default="canonical")
    p.add_argument("--workers", type=int, default=1,
                   help="CPU hogwild threads (numPartitions analog)")
    p.add_argument("--shards", type=int, default=1,
                   help="checkpoint shard files (numParameterServers analog)")
    p.add_argument("--resume-from", default=None, metavar="CKPT",
                   help="continue training from a saved model directory")
    p.add_argument("--checkpoint-every", type=int, default=0, metavar="N",
                   help="write a loadable checkpoint every N steps "
                        "(<model>-step<N> directories; 0 = off)")
    p.add_argument("--updates", choices=["hogwild", "atomic", "hybrid"],
                   default="hybrid",
                   help="row-update mode (hybrid: atomics on the hot "
                        "Zipf band, hogwild elsewhere — the measured "
                        "quality/speed default)")
    p.add_argument("--shared-negatives", action="store_true",
                   help="one negative set per position (HogBatch-style; "
                        "4.6x at high negative counts, see results.md)")


def main(argv=None):
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(name)s %(message)s")
    ap = argparse.ArgumentParser(prog="glint_word2vec_amd")
    sub = ap.add_subparsers(dest="cmd", required=True)
    _add_train(sub)
    p = sub.add_parser("similar", help="top-N cosine-similar words")
    p.add_argument("model"); p.add_argument("word")
    p.add_argument("-n", type=int, default=10)
    p.add_argument("--sharded", action="store_true",
                   help="stream the model to the GPU(s) shard-wise "
                        "(no host materialisation; torchrun for multi-GPU)")
    p = sub.add_parser("analogy", help="pos... - neg... vector arithmetic")
    p.add_argument("model"); p.add_argument("pos", nargs="+")
    p.add_argument("-m", "--minus", nargs="+", default=[])
    p.add_argument("-n", type=int, default=10)
    p.add_argument("--sharded", action="store_true")
    p = sub.add_parser("export", help="write word2vec text format (toLocal)")
    p.add_argument("model"); p.add_argument("out")
    p.add_argument("--sharded", action="store_true",
                   help="stream the export from device shards (no host "
                        "matrix; torchrun for multi-GPU)")
    p = sub.add_parser("info", help="print model metadata")
    p.add_argument("model")
    p = sub.add_parser("serve", help="HTTP serving endpoint (FastAPI)")
    p.add_argument("model")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--sharded", action="store_true",
                   help="stream the model to device shards (torchrun for "
                        "multi-GPU; every rank runs the app)")
    p.add_argument("--device", default=None)
    args = ap.parse_args(argv)

    if args.cmd == "train":
        est = GlintWord2Vec(
            vector_size=args.vector_size, learning_rate=args.learning_rate,
            num_iterations=args.iterations, window=args.window,
            n=args.negatives, min_count=args.min_count,
            subsample_ratio=args.subsample, seed=args.seed,
            unigram_table_size=args.unigram_table_size,
            max_sentence_length=args.max_sentence_length,
            dtype=args.dtype, device=args.device, engine=args.engine,
            window_mode=args.window_mode, num_partitions=args.workers,
            checkpoint_every=args.checkpoint_every,
            update_mode=args.updates,
            shared_negatives=args.shared_negatives)
        model = est.fit(args.corpus,
                        save_path=(args.model if args.checkpoint_every
                                   else None),
                        init_from=args.resume_from)
        import os
        rank = int(os.environ.get("RANK", "0"))
        if rank == 0:
            model.save(args.model, num_shards=args.shards)
            print(f"model saved to {args.model} "
                  f"({model.num_words} words, dim {model.vector_size})")
        model.stop()
    elif args.cmd == "similar":
        model = (GlintWord2VecModel.load_sharded(args.model) if args.sharded
                 else GlintWord2VecModel.load(args.model))
        for w, c in model.find_synonyms(args.word, args.n):
            print(f"{c:.4f}\t{w}")
    elif args.cmd == "analogy":
        model = (GlintWord2VecModel.load_sharded(args.model) if args.sharded
                 else GlintWord2VecModel.load(args.model))
        for w, c in model.analogy(args.pos, args.minus, args.n):
            print(f"{c:.4f}\t{w}")
    elif args.cmd == "export":
        if args.sharded:
            model = GlintWord2VecModel.load_sharded(args.model)
            model.export_text(args.out)
        else:
            model = GlintWord2VecModel.load(args.model)
            model.to_local().save(args.out)
        print(f"wrote {args.out}")
    elif args.cmd == "serve":
        from .server import serve
        serve(args.model, host=args.host, port=args.port,
              sharded=args.sharded, device=args.device)
    elif args.cmd == "info":
        import json
        import os
        meta = json.load(open(os.path.join(args.model, "metadata")))
        print(json.dumps(meta, indent=2))
    return 0


if __name__ == "__main__":
    sys.exit(main())
