"""EarlyStopping + same_seeds semantics (reference utils/customized.py)."""
import torch

from ddp_tricks_amd import EarlyStopping, same_seeds


def test_early_stopping_patience():
    es = EarlyStopping(patience=3, verbose=False)
    es(1.0)               # sets best
    for _ in range(2):
        es(2.0)           # worse
    assert not es.early_stop
    es(2.0)
    assert es.early_stop


def test_early_stopping_equal_is_improvement_at_delta0():
    es = EarlyStopping(patience=1, verbose=False)
    es(1.0)
    es(1.0)               # equal loss -> else branch -> counter reset
    assert es.counter == 0 and not es.early_stop


def test_early_stopping_improvement_resets():
    es = EarlyStopping(patience=2, verbose=False)
    es(1.0)
    es(1.5)
    assert es.counter == 1
    es(0.5)
    assert es.counter == 0
    es(0.9)
    assert es.counter == 1
    es(0.9)  # best is still 0.5, so another bad epoch -> patience reached
    assert es.early_stop


def test_early_stopping_metric_mode():
    es = EarlyStopping(patience=1, verbose=False)
    es(0.9, loss=False)
    es(0.95, loss=False)
    assert es.counter == 0
    es(0.8, loss=False)
    assert es.early_stop


def test_same_seeds_reproducible():
    same_seeds(7)
    a = torch.randn(4)
    same_seeds(7)
    b = torch.randn(4)
    assert torch.equal(a, b)
