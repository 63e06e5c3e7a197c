"""Flag system — CLI-compatible with the reference's absl flags.

The reference defines 15 absl flags (reference utils.py:17-33) plus
``--num_gpu`` (reference distributed_train.py:23) and collects them into a
kwargs dict with ``flags_dict()`` (reference utils.py:36-62).  absl is not
installed in this environment, so the same flag names, defaults and help
strings are provided through argparse; ``--flag value`` and ``--flag=value``
both work, and booleans additionally accept absl's ``--noenable_function``
negative form.
"""

from __future__ import annotations

import argparse
import sys


def _add_bool_flag(parser: argparse.ArgumentParser, name: str, default: bool, help_: str):
    """absl-style boolean: --name / --noname / --name=true|false."""

    class _BoolAction(argparse.Action):
        def __call__(self, p, ns, values, option_string=None):
            if option_string and option_string.startswith("--no"):
                setattr(ns, self.dest, False)
            elif values is None or values == []:
                setattr(ns, self.dest, True)
            else:
                v = values if isinstance(values, str) else values[0]
                setattr(ns, self.dest, str(v).lower() in ("1", "true", "t", "yes", "y"))

    parser.add_argument(f"--{name}", f"--no{name}", dest=name, action=_BoolAction,
                        nargs="?", default=default, help=help_)


def transformer_flags(parser: argparse.ArgumentParser | None = None) -> argparse.ArgumentParser:
    """Define the training flags (parity with reference utils.py:17-33).

    Extra flags beyond the reference (warmup_steps, label_smoothing, device,
    seed, log_interval, synthetic_*) are additive capabilities; their defaults
    reproduce reference behaviour.
    """
    if parser is None:
        parser = argparse.ArgumentParser(description="transformer_amd trainer")
    parser.add_argument("--dataset_path", type=str, default="data/", help=" Dataset Folder")
    parser.add_argument("--buffer_size", type=int, default=100000, help="Shuffle buffer size")
    parser.add_argument("--src_vocab_file", type=str, default="src_vocab.txt",
                        help="Source Vocabulary file")
    parser.add_argument("--tgt_vocab_file", type=str, default="tgt_vocab.txt",
                        help="Target Vocabulary file")
    parser.add_argument("--sequence_length", type=int, default=50,
                        help="Maxinum number of words in a sequence")
    parser.add_argument("--epochs", type=int, default=4, help="Number of Epochs")
    parser.add_argument("--batch_size", type=int, default=64, help="Batch Size")
    parser.add_argument("--per_replica_batch_size", type=int, default=16, help="Batch Size")
    parser.add_argument("--num_layers", type=int, default=4,
                        help="Nnmber of Encoder/Decoder Stack")
    parser.add_argument("--d_model", type=int, default=512,
                        help="Output dimesion of all sublayers including Embedding layer")
    parser.add_argument("--dff", type=int, default=1024, help="Dimetionality of inner layer")
    parser.add_argument("--num_heads", type=int, default=4, help="Number of Attention Head")
    parser.add_argument("--trace_dir", type=str, default=None,
                        help="If set, profile 5 training steps with torch.profiler "
                             "and write a chrome trace here before training")
    _add_bool_flag(parser, "enable_function", True,
                   "Enable Function (compile/capture the train step)")
    parser.add_argument("--max_ckpt_keep", type=int, default=5,
                        help="Maximum Number of Checkpoint to keep")
    parser.add_argument("--ckpt_path", type=str, default="model_dist", help="Checkpoint Path")
    parser.add_argument("--dropout_rate", type=float, default=0.1, help="Dropout Probability")
    # --- additive flags (defaults keep reference semantics) ---
    parser.add_argument("--warmup_steps", type=int, default=60000,
                        help="Noam schedule warmup steps (reference train.py:22 default)")
    parser.add_argument("--label_smoothing", type=float, default=0.0,
                        help="CE label smoothing; 0 reproduces reference plain CE "
                             "(SURVEY.md §8 Q9)")
    parser.add_argument("--device", type=str, default=None,
                        help="cuda|cpu; default auto-detect")
    parser.add_argument("--dtype", type=str, default=None,
                        help="bf16|fp32 compute dtype; default bf16 on GPU, fp32 on CPU")
    parser.add_argument("--seed", type=int, default=1234, help="RNG seed")
    parser.add_argument("--log_interval", type=int, default=100,
                        help="steps between test-eval/report (reference train.py:193)")
    parser.add_argument("--eval_steps", type=int, default=50,
                        help="number of test batches per eval (SURVEY.md §8 Q8 fix)")
    parser.add_argument("--synthetic_data", action="store_true",
                        help="use synthetic random-token data of the benchmark shape")
    parser.add_argument("--synthetic_vocab", type=int, default=32768,
                        help="token vocab for --synthetic_data (model vocab = +2)")
    parser.add_argument("--steps_per_epoch", type=int, default=100,
                        help="steps per epoch when --synthetic_data")
    parser.add_argument("--debug_sync", action="store_true",
                        help="serialize+block every kernel launch "
                             "(AMD_SERIALIZE_KERNEL=3, HIP_LAUNCH_BLOCKING=1)"
                             " — race/fault localization mode, SURVEY.md §5")
    parser.add_argument("--max_decode_len", type=int, default=10,
                        help="greedy decode steps for predict (reference train.py:109 uses 10)")
    return parser


def flags_dict(args: argparse.Namespace) -> dict:
    """Marshal parsed flags into the kwargs dict the mains consume
    (parity with reference utils.py:36-62, which drops per_replica_batch_size
    for the single-device main — kept here since our main ignores extras)."""
    return {
        "dataset_path": args.dataset_path,
        "enable_function": args.enable_function,
        "buffer_size": args.buffer_size,
        "src_vocab_file": args.src_vocab_file,
        "tgt_vocab_file": args.tgt_vocab_file,
        "batch_size": args.batch_size,
        "per_replica_batch_size": args.per_replica_batch_size,
        "sequence_length": args.sequence_length,
        "epochs": args.epochs,
        "num_layers": args.num_layers,
        "d_model": args.d_model,
        "dff": args.dff,
        "num_heads": args.num_heads,
        "max_ckpt_keep": args.max_ckpt_keep,
        "ckpt_path": args.ckpt_path,
        "dropout_rate": args.dropout_rate,
        "warmup_steps": args.warmup_steps,
        "label_smoothing": args.label_smoothing,
        "device": args.device,
        "dtype": args.dtype,
        "seed": args.seed,
        "log_interval": args.log_interval,
        "eval_steps": args.eval_steps,
        "synthetic_data": args.synthetic_data,
        "synthetic_vocab": args.synthetic_vocab,
        "steps_per_epoch": args.steps_per_epoch,
        "max_decode_len": args.max_decode_len,
        "debug_sync": args.debug_sync,
    }


def parse_flags(argv=None, extra=None):
    parser = transformer_flags()
    if extra:
        extra(parser)
    args, unknown = parser.parse_known_args(argv if argv is not None else sys.argv[1:])
    if unknown:
        print(f"[transformer_amd] ignoring unknown flags: {unknown}", file=sys.stderr)
    return args
