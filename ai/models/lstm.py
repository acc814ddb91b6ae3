"""Shim: the reference layout's ai/models/lstm.py -> nerrf_amd."""
from nerrf_amd.models.lstm import BiLSTMDetector, FusedLSTMDirection, LSTMConfig  # noqa: F401
