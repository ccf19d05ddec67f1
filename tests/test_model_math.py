"""Model/op math parity with the reference semantics (transcribed oracle).

The eager ops in zaremba_amd.ops.functional must reproduce the exact math
of the reference cell (model.py:34-45), loss (main.py:77-84) and SGD step
(main.py:115-117), including the i,f,o,n gate order and the batch_size
loss scaling.
"""

import math

import pytest
import torch

from zaremba_amd.models.lstm_lm import Model
from zaremba_amd.ops import functional as F_ref


def _naive_reference_step(x, h, c, W_x, W_h, b_x, b_h):
    """Literal transcription of the reference lstm_step for testing."""
    gx = torch.addmm(b_x, x, W_x.t())
    gh = torch.addmm(b_h, h, W_h.t())
    xi, xf, xo, xn = gx.chunk(4, 1)
    hi, hf, ho, hn = gh.chunk(4, 1)
    i = torch.sigmoid(xi + hi)
    f = torch.sigmoid(xf + hf)
    o = torch.sigmoid(xo + ho)
    n = torch.tanh(xn + hn)
    c2 = f * c + i * n
    h2 = o * torch.tanh(c2)
    return h2, c2


def test_lstm_step_matches_reference_math():
    torch.manual_seed(0)
    B, H, X = 5, 16, 16
    x = torch.randn(B, X)
    h = torch.randn(B, H)
    c = torch.randn(B, H)
    W_x = torch.randn(4 * H, X) * 0.1
    W_h = torch.randn(4 * H, H) * 0.1
    b_x = torch.randn(4 * H) * 0.1
    b_h = torch.randn(4 * H) * 0.1
    h1, c1 = F_ref.lstm_step(x, h, c, W_x, W_h, b_x, b_h)
    h2, c2 = _naive_reference_step(x, h, c, W_x, W_h, b_x, b_h)
    assert torch.equal(h1, h2) and torch.equal(c1, c2)


def test_gate_order_differs_from_nn_lstm():
    """The i,f,o,n order must NOT match nn.LSTM's i,f,g,o (documented
    incompatibility, SURVEY.md component 11)."""
    torch.manual_seed(1)
    H = 8
    cell = torch.nn.LSTMCell(H, H)
    x = torch.randn(3, H)
    h = torch.randn(3, H)
    c = torch.randn(3, H)
    ours, _ = F_ref.lstm_step(x, h, c, cell.weight_ih, cell.weight_hh,
                              cell.bias_ih, cell.bias_hh)
    theirs, _ = cell(x, (h, c))
    assert not torch.allclose(ours, theirs)


def test_nll_loss_matches_naive_softmax():
    """Stable log-softmax NLL == the reference's naive exp/normalize math
    (main.py:77-84) wherever the naive form doesn't overflow."""
    torch.manual_seed(2)
    T, B, V = 4, 3, 50
    scores = torch.randn(T * B, V) * 3
    y = torch.randint(0, V, (T, B))
    # naive reference math
    expscores = scores.exp()
    probs = expscores / expscores.sum(1, keepdim=True)
    ans = probs[range(y.numel()), y.reshape(-1)]
    naive = torch.mean(-torch.log(ans) * B)
    ours = F_ref.nll_loss(scores, y)
    assert torch.allclose(ours, naive, atol=1e-5)


def test_nll_loss_stable_at_large_logits():
    scores = torch.full((6, 10), 500.0)
    y = torch.zeros(2, 3, dtype=torch.int64)
    loss = F_ref.nll_loss(scores, y)
    assert torch.isfinite(loss)
    assert torch.allclose(loss, torch.tensor(math.log(10.0) * 3))


def test_clip_and_sgd_matches_torch():
    torch.manual_seed(3)
    mk = lambda: torch.nn.Parameter(torch.randn(7, 5))
    a1, b1 = mk(), mk()
    a2 = torch.nn.Parameter(a1.detach().clone())
    b2 = torch.nn.Parameter(b1.detach().clone())
    g1, g2 = torch.randn(7, 5) * 10, torch.randn(7, 5) * 10
    for p, g in ((a1, g1), (b1, g2), (a2, g1), (b2, g2)):
        p.grad = g.clone()
    norm = F_ref.clip_grad_and_sgd_([a1, b1], max_norm=1.0, lr=0.5)
    expected_norm = torch.nn.utils.clip_grad_norm_([a2, b2], 1.0)
    with torch.no_grad():
        for p in (a2, b2):
            p -= 0.5 * p.grad
    assert torch.allclose(norm, expected_norm)
    assert torch.allclose(a1, a2, atol=1e-6)
    assert torch.allclose(b1, b2, atol=1e-6)


def test_model_forward_shapes_and_state():
    torch.manual_seed(4)
    V, H, L, B, T = 37, 12, 2, 5, 6
    model = Model(V, H, L, dropout=0.0, winit=0.1, lstm_type="custom")
    states = model.state_init(B)
    assert len(states) == L and states[0][0].shape == (B, H)
    x = torch.randint(0, V, (T, B))
    scores, states = model(x, states)
    assert scores.shape == (T * B, V)
    assert states[0][0].shape == (B, H)
    # state actually evolves
    assert not torch.equal(states[0][0], torch.zeros(B, H))


def test_model_param_names_match_reference_convention():
    model = Model(10, 4, 2, 0.0, 0.1)
    names = {n for n, _ in model.named_parameters()}
    assert "embed.W" in names
    assert "rnns.0.W_x" in names and "rnns.1.b_h" in names
    assert "fc.W" in names and "fc.b" in names
    # exactly 1 (embed) + 4*L (rnns) + 2 (fc) parameters
    assert len(names) == 1 + 4 * 2 + 2


def test_model_init_uniform_winit():
    model = Model(50, 8, 1, 0.0, winit=0.07)
    for p in model.parameters():
        assert p.abs().max().item() <= 0.07 + 1e-6


def test_dropout_fresh_mask_per_call():
    model = Model(10, 64, 1, dropout=0.5, winit=0.1)
    model.train()
    x = torch.ones(4, 3, 64)
    a = model.dropout(x)
    b = model.dropout(x)
    assert not torch.equal(a, b)
    model.eval()
    assert torch.equal(model.dropout(x), x)


def test_lstm_layer_gradcheck_fp64():
    """fp64 torch.autograd.gradcheck of the eager oracle cell (SURVEY §4:
    the oracle every HIP kernel is validated against must itself have
    exact gradients)."""
    import torch
    from zaremba_amd.ops import functional as F_ref

    torch.manual_seed(0)
    T, B, H = 3, 2, 4
    x = torch.randn(T, B, H, dtype=torch.float64, requires_grad=True)
    h0 = torch.randn(B, H, dtype=torch.float64, requires_grad=True)
    c0 = torch.randn(B, H, dtype=torch.float64, requires_grad=True)
    Wx = torch.randn(4 * H, H, dtype=torch.float64, requires_grad=True)
    Wh = torch.randn(4 * H, H, dtype=torch.float64, requires_grad=True)
    bx = torch.randn(4 * H, dtype=torch.float64, requires_grad=True)
    bh = torch.randn(4 * H, dtype=torch.float64, requires_grad=True)

    def f(x, h0, c0, Wx, Wh, bx, bh):
        out, h, c = F_ref.lstm_layer(x, h0, c0, Wx, Wh, bx, bh)
        return out.sum() + h.sum() + c.sum()

    assert torch.autograd.gradcheck(f, (x, h0, c0, Wx, Wh, bx, bh),
                                    eps=1e-6, atol=1e-8)
