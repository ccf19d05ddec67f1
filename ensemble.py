#!/usr/bin/env python3
"""Ensemble (model-averaging) training — drop-in CLI for the reference's
ensemble.py (same flags/defaults, reference ensemble.py:9-42), rebuilt for
MI355X.

Single process: trains ensemble_num fresh models sequentially and reports
the incremental k-model averaged perplexity after each, exactly like the
reference (ensemble.py:166-182) — but also persists every model (the
reference kept them only in memory).

Multi-GPU (launch via torchrun, one rank per GPU): models are trained in
parallel round-robin across ranks (model i on rank i % world), each saved
to --save_dir and kept resident on its trainer rank; the incremental
ensemble perplexities are then evaluated DISTRIBUTED — every rank forwards
its own members and one RCCL all-reduce(SUM) per batch merges the
probability sums (reference math ensemble.py:97-126; BASELINE config 5).
"""

import argparse
import os

import torch

from zaremba_amd import data as zdata
from zaremba_amd import trainer
from zaremba_amd.checkpoint import save_checkpoint
from zaremba_amd.ensemble_eval import (ensemble_perplexity,
                                       ensemble_perplexity_distributed)
from zaremba_amd.models.lstm_lm import Model
from zaremba_amd.parallel import dist as zdist

from main import setdevice  # same device policy/messages


def build_parser():
    parser = argparse.ArgumentParser(
        description="Replication of Zaremba et al. (2014). \n"
                    "https://arxiv.org/abs/1409.2329")
    parser.add_argument("--ensemble_num", type=int, default=5,
                        help="The number of models to average.")
    parser.add_argument("--layer_num", type=int, default=2,
                        help="The number of LSTM layers the model has.")
    parser.add_argument("--hidden_size", type=int, default=200,
                        help="The number of hidden units per layer.")
    parser.add_argument("--lstm_type", type=str, choices=["pytorch", "custom"],
                        default="pytorch",
                        help="Which implementation of LSTM to use. Both map to "
                             "the fused MI355X HIP cell in this framework.")
    parser.add_argument("--dropout", type=float, default=0.0,
                        help="The dropout parameter.")
    parser.add_argument("--winit", type=float, default=0.1,
                        help="The weight initialization parameter.")
    parser.add_argument("--batch_size", type=int, default=20,
                        help="The batch size.")
    parser.add_argument("--seq_length", type=int, default=20,
                        help="The sequence length for bptt.")
    parser.add_argument("--learning_rate", type=float, default=1,
                        help="The learning rate.")
    parser.add_argument("--total_epochs", type=int, default=13,
                        help="Total number of epochs for training.")
    parser.add_argument("--factor_epoch", type=int, default=4,
                        help="The epoch to start factoring the learning rate.")
    parser.add_argument("--factor", type=float, default=2,
                        help="The factor to decrease the learning rate.")
    parser.add_argument("--max_grad_norm", type=float, default=5,
                        help="The maximum norm of gradients we impose on training.")
    parser.add_argument("--device", type=str, choices=["cpu", "gpu"],
                        default="gpu",
                        help="Whether to use cpu or gpu. On default falls back "
                             "to gpu if one exists, falls back to cpu otherwise.")
    # Additive flags.
    parser.add_argument("--seed", type=int, default=None)
    parser.add_argument("--engine", type=str,
                        choices=["auto", "hip", "eager"], default="auto")
    parser.add_argument("--dtype", type=str, choices=["fp32", "bf16"],
                        default="bf16")
    parser.add_argument("--data", type=str, default="ptb")
    parser.add_argument("--data_dir", type=str, default="./data")
    parser.add_argument("--save_dir", type=str, default=None,
                        help="Directory for per-model checkpoints "
                             "(required in multi-rank mode).")
    parser.add_argument("--jsonl", type=str, default=None)
    return parser


def load_data(args):
    from main import load_data as _ld
    return _ld(args)


def train_one(args, vocab_size, data, model_num, jsonl_path=None):
    if args.seed is not None:
        torch.manual_seed(args.seed + 1000 * model_num)
    engine = args.engine
    if args.device.type == "cuda" and args.dtype == "fp32":
        if engine == "hip":
            raise SystemExit(
                "--dtype fp32 is not supported by --engine hip (bf16 "
                "compute / fp32 masters); use --engine eager or auto.")
        engine = "eager"
        os.environ["ZAREMBA_AMD_FORCE_EAGER"] = "1"
    model = Model(vocab_size, args.hidden_size, args.layer_num, args.dropout,
                  args.winit, args.lstm_type, engine=engine)
    model.to(args.device)
    if args.device.type == "cuda" and args.dtype == "bf16":
        model.hip().set_compute_dtype(torch.bfloat16)
    trainer.train(data, model, args.total_epochs, args.factor_epoch,
                  args.learning_rate, args.factor, args.max_grad_norm,
                  args.batch_size, log_every=800, model_num=model_num,
                  jsonl_path=jsonl_path, is_rank0=True)
    return model


def jsonl_for_model(args, model_num):
    """Per-model JSONL path. In multi-rank mode every rank logs with
    is_rank0=True, so a shared file would interleave concurrent appends
    (advisor finding); suffix with the model number instead."""
    if args.jsonl is None:
        return None
    if zdist.world_size() == 1:
        return args.jsonl
    root, ext = os.path.splitext(args.jsonl)
    return f"{root}.model{model_num}{ext or '.jsonl'}"


def main():
    args = build_parser().parse_args()
    if args.engine == "hip" and args.dtype == "fp32":
        raise SystemExit(
            "--dtype fp32 is not supported by --engine hip (bf16 compute / "
            "fp32 masters); use --engine eager or auto.")
    if args.engine == "eager":
        os.environ["ZAREMBA_AMD_FORCE_EAGER"] = "1"
    zdist.maybe_init()
    setdevice(args, plural=True)
    if zdist.is_rank0():
        print("Parameters for the base model of the ensemble:")
        print("Args:", args)
        print("\n")

    trn_raw, vld_raw, tst_raw, vocab_size = load_data(args)
    trn = zdata.minibatch(trn_raw, args.batch_size, args.seq_length)
    vld = zdata.minibatch(vld_raw, args.batch_size, args.seq_length)
    tst = zdata.minibatch(tst_raw, args.batch_size, args.seq_length)
    data = (trn, vld, tst)

    world = zdist.world_size()
    if world > 1:
        if not args.save_dir:
            raise SystemExit("--save_dir is required for multi-rank ensemble "
                             "training")
        os.makedirs(args.save_dir, exist_ok=True)
        # Parallel: model i trains on rank i % world (diversity = seeds/init)
        # and STAYS resident on its trainer rank for the distributed eval.
        my_models = {}
        for i in range(args.ensemble_num):
            if i % world == zdist.rank():
                model = train_one(args, vocab_size, data, i + 1,
                                  jsonl_path=jsonl_for_model(args, i + 1))
                save_checkpoint(os.path.join(args.save_dir, f"model_{i + 1}.pt"),
                                model, epoch=args.total_epochs,
                                lr=args.learning_rate)
                my_models[i + 1] = model
        zdist.barrier()
        # Distributed incremental eval (BASELINE config 5 / SURVEY K13):
        # for each ensemble size k, every rank forwards only the members it
        # trained; one all-reduce(SUM) per batch merges the prob sums over
        # RCCL/xGMI. Ranks owning no member <= k contribute zeros.
        for k in range(1, args.ensemble_num + 1):
            subset = {f"model {j}": m for j, m in my_models.items() if j <= k}
            val_perp = ensemble_perplexity_distributed(
                vld, subset, k, args.batch_size, vocab_size, args.device)
            tst_perp = ensemble_perplexity_distributed(
                tst, subset, k, args.batch_size, vocab_size, args.device)
            if zdist.is_rank0():
                print("Validation set perplexity of {} averaged models: {:.3f}"
                      .format(k, val_perp))
                print("Test set perplexity of {} averaged models: {:.3f}\n"
                      .format(k, tst_perp))
        zdist.finalize()
        return

    # Single process: the reference's sequential flow (ensemble.py:166-182).
    models = {}
    for i in range(args.ensemble_num):
        model = train_one(args, vocab_size, data, i + 1,
                          jsonl_path=jsonl_for_model(args, i + 1))
        models["model {:d}".format(i + 1)] = model
        if args.save_dir:
            os.makedirs(args.save_dir, exist_ok=True)
            save_checkpoint(os.path.join(args.save_dir, f"model_{i + 1}.pt"),
                            model, epoch=args.total_epochs, lr=args.learning_rate)
        val_perp = ensemble_perplexity(vld, models, args.batch_size)
        print("Validation set perplexity of {} averaged models: {:.3f}".format(
            i + 1, val_perp))
        tst_perp = ensemble_perplexity(tst, models, args.batch_size)
        print("Test set perplexity of {} averaged models: {:.3f}\n".format(
            i + 1, tst_perp))


if __name__ == "__main__":
    main()
