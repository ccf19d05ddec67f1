"""End-to-end convergence validation on the learnable Markov corpus:
the HIP bf16 path must reach near-optimal perplexity (= branch factor).
Usage: python tools/convergence_check.py [epochs] [engine]"""
import sys, os, io, contextlib
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from zaremba_amd import data as zdata, trainer
from zaremba_amd.models.lstm_lm import Model

epochs = int(sys.argv[1]) if len(sys.argv) > 1 else 12
engine = sys.argv[2] if len(sys.argv) > 2 else "auto"
vocab = int(sys.argv[3]) if len(sys.argv) > 3 else 2000
# ~50 observations per (state, successor) transition: learnable in a few
# epochs (vocab 10000 with only 400K tokens gave ~2 obs/transition and
# plateaued at the unigram entropy for EVERY engine, including eager fp32)
tokens = vocab * 20 * 50
torch.manual_seed(0)
trn, vld, tst, v = zdata.synthetic_markov_init(
    vocab_size=vocab, branch=20, train_tokens=tokens,
    valid_tokens=tokens // 10, test_tokens=tokens // 10, seed=1)
ds = zdata.minibatch(trn, 20, 35)
dv = zdata.minibatch(vld, 20, 35)
dev = "cuda" if torch.cuda.is_available() else "cpu"
m = Model(v, 650, 2, dropout=0.0, winit=0.05, engine=engine).to(dev)
for ep in range(epochs):
    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        trainer.train((ds, dv, dv), m, epochs=1, epoch_threshold=100, lr=1.0,
                      factor=1.2, max_norm=5.0, batch_size=20)
    print(f"epoch {ep+1}: valid ppl = "
          f"{trainer.perplexity(dv, m, 20):.2f}", flush=True)
print("optimal ppl = 20 (branch factor)")
