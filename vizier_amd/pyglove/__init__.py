"""Public PyGlove integration (parity with vizier/pyglove).

PyGlove itself is an optional dependency; attributes that need it are
resolved lazily so this module imports cleanly without it.
"""

from vizier_amd._src.pyglove.integration import (
    VizierConverter,
    create_policy,
    init,
)


def __getattr__(name):
  if name == 'BuiltinAlgorithm':
    from vizier_amd._src.pyglove.integration import (
        make_builtin_algorithm_class,
    )
    return make_builtin_algorithm_class()
  raise AttributeError(name)
