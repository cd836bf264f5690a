"""HEBO-style alternative GP (vizier_amd/_src/gp/hebo.py)."""

import math

import numpy as np
import pytest
import torch

from vizier_amd._src.gp import hebo


def make_data(n=30, d=3, seed=0):
  g = torch.Generator().manual_seed(seed)
  x = torch.rand(n, d, generator=g)
  y = torch.sin(4 * x[:, 0]) - x[:, 1] ** 2 + \
      0.05 * torch.randn(n, generator=g)
  return x, y


class TestPieces:

  def test_kumaraswamy_warp_monotone_and_bounded(self):
    x = torch.linspace(0.01, 0.99, 50).reshape(-1, 1)
    w = hebo.kumaraswamy_warp(x, torch.tensor(2.0), torch.tensor(0.7))
    assert (w >= 0).all() and (w <= 1).all()
    assert (w[1:] > w[:-1]).all()
    # Identity at c0 = c1 = 1.
    w_id = hebo.kumaraswamy_warp(x, torch.tensor(1.0), torch.tensor(1.0))
    assert torch.allclose(w_id, x.clamp(1e-6, 1 - 1e-6), atol=1e-6)

  def test_matern32_values(self):
    assert float(hebo.matern32(torch.tensor(0.0))) == pytest.approx(1.0)
    r = torch.tensor(1.0)
    want = (1 + math.sqrt(3)) * math.exp(-math.sqrt(3))
    assert float(hebo.matern32(r)) == pytest.approx(want, rel=1e-6)

  def test_gram_psd_and_linear_term(self):
    x, _ = make_data()
    raw = torch.zeros(1, x.shape[1] + 4)
    params = hebo.HeboParams.from_raw(raw, x.shape[1])
    K = hebo._hebo_gram(params, x.unsqueeze(0), None)[0]
    assert torch.allclose(K, K.T, atol=1e-6)
    ev = torch.linalg.eigvalsh(K.double())
    assert float(ev.min()) > -1e-8

  def test_nlp_finite_and_prior_pulls(self):
    x, y = make_data()
    raw = torch.randn(4, x.shape[1] + 4) * 0.5
    nlp = hebo.negative_log_posterior(raw, x, (y - y.mean()) / y.std())
    assert torch.isfinite(nlp).all()


class TestTraining:

  def test_fit_interpolates(self):
    x, y = make_data(n=40)
    post = hebo.train_hebo_gp(x, y, num_restarts=2, max_iters=25)
    mean, stddev = post.predict(x)
    assert float((mean - y).abs().mean()) < 0.25
    assert (stddev > 0).all()

  def test_predictions_generalize(self):
    x, y = make_data(n=60, seed=1)
    post = hebo.train_hebo_gp(x, y, num_restarts=2, max_iters=25)
    g = torch.Generator().manual_seed(9)
    xq = torch.rand(30, 3, generator=g)
    f = torch.sin(4 * xq[:, 0]) - xq[:, 1] ** 2
    mean, _ = post.predict(xq)
    # Better than predicting the training mean.
    base = float((y.mean() - f).abs().mean())
    got = float((mean - f).abs().mean())
    assert got < base, (got, base)

  def test_label_standardization_roundtrip(self):
    x, y = make_data(n=30)
    y_shifted = y * 50.0 + 300.0
    post = hebo.train_hebo_gp(x, y_shifted, num_restarts=1, max_iters=15)
    mean, _ = post.predict(x)
    assert float((mean - y_shifted).abs().mean()) < 15.0
