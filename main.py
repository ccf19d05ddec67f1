#!/usr/bin/env python3
"""Train the Zaremba LSTM language model — drop-in CLI for the reference's
main.py (same flags, defaults and console output; reference main.py:10-41),
running on the MI355X-native zaremba_amd framework.

Additive flags (absent from the reference): --seed, --engine, --dtype,
--data, --data_dir, --save, --resume, --jsonl. Multi-GPU data parallelism:
launch with `python -m torch.distributed.run --nproc-per-node N
--master-addr 127.0.0.1 main.py ...` — one rank per GPU over RCCL/xGMI.
"""

import argparse
import os

import torch

from zaremba_amd import data as zdata
from zaremba_amd import trainer
from zaremba_amd.models.lstm_lm import Model
from zaremba_amd.parallel import dist as zdist


def build_parser():
    parser = argparse.ArgumentParser(
        description="Replication of Zaremba et al. (2014). \n"
                    "https://arxiv.org/abs/1409.2329")
    # Reference flag set (main.py:11-25), identical names/defaults/help intent.
    parser.add_argument("--layer_num", type=int, default=2,
                        help="The number of LSTM layers the model has.")
    parser.add_argument("--hidden_size", type=int, default=650,
                        help="The number of hidden units per layer.")
    parser.add_argument("--lstm_type", type=str, choices=["pytorch", "custom"],
                        default="pytorch",
                        help="Which implementation of LSTM to use. Both map to "
                             "the fused MI355X HIP cell in this framework.")
    parser.add_argument("--dropout", type=float, default=0.5,
                        help="The dropout parameter.")
    parser.add_argument("--winit", type=float, default=0.05,
                        help="The weight initialization parameter.")
    parser.add_argument("--batch_size", type=int, default=20,
                        help="The batch size (per GPU under data parallelism).")
    parser.add_argument("--seq_length", type=int, default=35,
                        help="The sequence length for bptt.")
    parser.add_argument("--learning_rate", type=float, default=1,
                        help="The learning rate.")
    parser.add_argument("--total_epochs", type=int, default=39,
                        help="Total number of epochs for training.")
    parser.add_argument("--factor_epoch", type=int, default=6,
                        help="The epoch to start factoring the learning rate.")
    parser.add_argument("--factor", type=float, default=1.2,
                        help="The factor to decrease the learning rate.")
    parser.add_argument("--max_grad_norm", type=float, default=5,
                        help="The maximum norm of gradients we impose on training.")
    parser.add_argument("--device", type=str, choices=["cpu", "gpu"],
                        default="gpu",
                        help="Whether to use cpu or gpu. On default falls back "
                             "to gpu if one exists, falls back to cpu otherwise.")
    # Additive flags.
    parser.add_argument("--seed", type=int, default=None,
                        help="Random seed (the reference has none).")
    parser.add_argument("--engine", type=str,
                        choices=["auto", "hip", "eager"], default="auto",
                        help="Compute path: fused HIP kernels (GPU) or eager "
                             "PyTorch ops (CPU / debugging).")
    parser.add_argument("--dtype", type=str, choices=["fp32", "bf16"],
                        default="bf16",
                        help="GPU compute dtype (fp32 master weights either way).")
    parser.add_argument("--data", type=str, default="ptb",
                        help="'ptb', 'synthetic[:vocab=N]' (uniform random, "
                             "PTB-shaped) or 'synthetic_markov[:vocab=N,"
                             "branch=K]' (learnable order-1 chain, optimal "
                             "perplexity = K).")
    parser.add_argument("--data_dir", type=str, default="./data",
                        help="Directory holding ptb.{train,valid,test}.txt.")
    parser.add_argument("--save", type=str, default=None,
                        help="Checkpoint path written after every epoch.")
    parser.add_argument("--resume", type=str, default=None,
                        help="Checkpoint path to resume from.")
    parser.add_argument("--jsonl", type=str, default=None,
                        help="Machine-readable JSONL mirror of the train log.")
    return parser


def setdevice(args, plural=False):
    """Device policy with the reference's exact fallback messages
    (main.py:28-39)."""
    subject = "Models" if plural else "Model"
    if args.device == "gpu" and torch.cuda.is_available():
        print(f"{subject} will be training on the GPU.\n")
        args.device = torch.device("cuda", zdist.local_rank())
    elif args.device == "gpu":
        print("No GPU detected. Falling back to CPU.\n")
        args.device = torch.device("cpu")
    else:
        print(f"{subject} will be training on the CPU.\n")
        args.device = torch.device("cpu")


def load_data(args):
    if args.data.startswith("synthetic"):
        opts = {"vocab": 10000, "branch": 20, "tokens": 929589}
        if ":" in args.data:
            for kv in args.data.split(":", 1)[1].split(","):
                k, v = kv.split("=")
                opts[k] = int(v)
        seed = args.seed if args.seed is not None else 1234
        sizes = dict(train_tokens=opts["tokens"],
                     valid_tokens=max(400, opts["tokens"] // 12),
                     test_tokens=max(400, opts["tokens"] // 11))
        if args.data.startswith("synthetic_markov"):
            return zdata.synthetic_markov_init(
                vocab_size=opts["vocab"], branch=opts["branch"], seed=seed,
                **sizes)
        return zdata.synthetic_init(vocab_size=opts["vocab"], seed=seed,
                                    **sizes)
    return zdata.data_init(args.data_dir)


def main():
    args = build_parser().parse_args()
    if args.engine == "hip" and args.dtype == "fp32":
        # The HIP kernel path is bf16-compute / fp32-master by design;
        # refuse up front rather than silently computing in bf16.
        raise SystemExit(
            "--dtype fp32 is not supported by --engine hip (the HIP "
            "kernels compute in bf16 with fp32 master weights). Use "
            "--engine eager (or auto) for full-fp32 compute.")
    if args.engine == "eager":
        os.environ["ZAREMBA_AMD_FORCE_EAGER"] = "1"
    zdist.maybe_init()
    if args.seed is not None:
        torch.manual_seed(args.seed + zdist.rank())
    setdevice(args)
    if zdist.is_rank0():
        print("Parameters of the model:")
        print("Args:", args)
        print("\n")

    trn_raw, vld_raw, tst_raw, vocab_size = load_data(args)
    if zdist.world_size() > 1:
        trn_raw = zdata.shard_stream(trn_raw, zdist.rank(), zdist.world_size())
    trn = zdata.minibatch(trn_raw, args.batch_size, args.seq_length)
    vld = zdata.minibatch(vld_raw, args.batch_size, args.seq_length)
    tst = zdata.minibatch(tst_raw, args.batch_size, args.seq_length)

    start_epoch, start_lr = 0, args.learning_rate
    if args.resume and os.path.exists(args.resume):
        from zaremba_amd.checkpoint import (build_model_from_checkpoint,
                                            restore_rng)
        model, payload = build_model_from_checkpoint(args.resume, engine=args.engine)
        start_epoch = payload["epoch"]
        # Restore the RNG streams so a resumed run's dropout masks and
        # any data draws continue the unbroken run's sequence.
        restore_rng(payload)
        if zdist.is_rank0():
            print(f"Resumed from {args.resume} at epoch {start_epoch}.")
    else:
        model = Model(vocab_size, args.hidden_size, args.layer_num,
                      args.dropout, args.winit, args.lstm_type,
                      engine=args.engine)
    if args.device.type == "cuda" and args.dtype == "fp32":
        # engine auto (hip+fp32 was rejected at parse time): fp32 compute
        # on GPU = the eager engine
        model.engine = "eager"
        os.environ["ZAREMBA_AMD_FORCE_EAGER"] = "1"
    model.to(args.device)
    if args.device.type == "cuda" and args.dtype == "bf16":
        model.hip().set_compute_dtype(torch.bfloat16)

    dp = None
    if zdist.world_size() > 1:
        zdist.broadcast_parameters(model)
        from zaremba_amd.parallel.bucketer import GradBucketer
        dp = GradBucketer(model)

    trainer.train(
        (trn, vld, tst), model, args.total_epochs, args.factor_epoch,
        start_lr, args.factor, args.max_grad_norm, args.batch_size,
        jsonl_path=args.jsonl, save_path=args.save, start_epoch=start_epoch,
        dp=dp, is_rank0=zdist.is_rank0(),
    )
    zdist.finalize()


if __name__ == "__main__":
    main()
