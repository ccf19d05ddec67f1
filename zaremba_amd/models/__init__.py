from .lstm_lm import Embed, LSTM, Linear, Model  # noqa: F401
