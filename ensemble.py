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
to --save_dir; rank 0 then reports the incremental ensemble perplexities
over the saved checkpoints.
"""

import argparse
import os

import torch

from zaremba_amd import data as zdata
from zaremba_amd import trainer
from zaremba_amd.checkpoint import build_model_from_checkpoint, save_checkpoint
from zaremba_amd.ensemble_eval import ensemble_perplexity
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


def train_one(args, vocab_size, data, model_num):
    if args.seed is not None:
        torch.manual_seed(args.seed + 1000 * model_num)
    engine = args.engine
    if args.device.type == "cuda" and args.dtype == "fp32" \
            and engine == "auto":
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
                  jsonl_path=args.jsonl, is_rank0=True)
    return model


def main():
    args = build_parser().parse_args()
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
        # Parallel: model i trains on rank i % world (diversity = seeds/init).
        for i in range(args.ensemble_num):
            if i % world == zdist.rank():
                model = train_one(args, vocab_size, data, i + 1)
                save_checkpoint(os.path.join(args.save_dir, f"model_{i + 1}.pt"),
                                model, epoch=args.total_epochs,
                                lr=args.learning_rate)
                del model
        zdist.barrier()
        if zdist.is_rank0():
            models = {}
            for i in range(args.ensemble_num):
                path = os.path.join(args.save_dir, f"model_{i + 1}.pt")
                m, _ = build_model_from_checkpoint(path, engine=args.engine)
                m.to(args.device)
                models[f"model {i + 1}"] = m
                val_perp = ensemble_perplexity(vld, models, args.batch_size)
                print("Validation set perplexity of {} averaged models: {:.3f}"
                      .format(i + 1, val_perp))
                tst_perp = ensemble_perplexity(tst, models, args.batch_size)
                print("Test set perplexity of {} averaged models: {:.3f}\n"
                      .format(i + 1, tst_perp))
        zdist.finalize()
        return

    # Single process: the reference's sequential flow (ensemble.py:166-182).
    models = {}
    for i in range(args.ensemble_num):
        model = train_one(args, vocab_size, data, i + 1)
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
