"""Shared pytest configuration."""

import pytest


def pytest_configure(config):
  config.addinivalue_line(
      'markers', 'gpu: tests that require an MI355X GPU (run via gpurun)')


def pytest_collection_modifyitems(config, items):
  try:
    import torch
    has_gpu = torch.cuda.is_available()
  except Exception:
    has_gpu = False
  if has_gpu:
    return
  skip_gpu = pytest.mark.skip(reason='no GPU available in this environment')
  for item in items:
    if 'gpu' in item.keywords:
      item.add_marker(skip_gpu)
