"""Public PyGlove integration (parity with vizier/pyglove).

Exposes the pg.tuning backend plugin surface of the reference
(`OSSVizierBackend`, `VizierBackend`, `TunerPolicy`, `VizierConverter`,
`BuiltinAlgorithm`, `init`, `create_policy`, `poll_result` via the
backend) plus the small legacy `VizierTuner` helper from round 1.

PyGlove itself is an optional dependency: every attribute that needs
it resolves lazily, so this module imports cleanly without pyglove and
raises an informative ImportError only on use.
"""

_PLUGIN_ATTRS = {
    'OSSVizierBackend': ('vizier_amd._src.pyglove.oss_vizier',
                         'OSSVizierBackend'),
    'init': ('vizier_amd._src.pyglove.oss_vizier', 'init'),
    'VizierBackend': ('vizier_amd._src.pyglove.backend', 'VizierBackend'),
    'TunerPolicy': ('vizier_amd._src.pyglove.tuner_policy', 'TunerPolicy'),
    'create_policy': ('vizier_amd._src.pyglove.tuner_policy',
                      'create_policy'),
    'VizierConverter': ('vizier_amd._src.pyglove.converters',
                        'VizierConverter'),
    'BuiltinAlgorithm': ('vizier_amd._src.pyglove.algorithms',
                         'BuiltinAlgorithm'),
    'Feedback': ('vizier_amd._src.pyglove.core', 'Feedback'),
    'VizierTrial': ('vizier_amd._src.pyglove.core', 'VizierTrial'),
    'Result': ('vizier_amd._src.pyglove.core', 'Result'),
    # Legacy round-1 helpers (small custom tuner API).
    'VizierTuner': ('vizier_amd._src.pyglove.vizier_backend',
                    'VizierTuner'),
}


def __getattr__(name):
  if name in _PLUGIN_ATTRS:
    import importlib
    module_name, attr = _PLUGIN_ATTRS[name]
    try:
      module = importlib.import_module(module_name)
    except ImportError as e:
      raise ImportError(
          f'vizier_amd.pyglove.{name} requires the `pyglove` package, '
          f'which is not installed: {e}') from e
    return getattr(module, attr)
  raise AttributeError(name)
