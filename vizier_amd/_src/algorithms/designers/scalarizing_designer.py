"""Multi-objective -> single-objective designer wrapper.

Capability parity with
vizier/_src/algorithms/designers/scalarizing_designer.py (:30,:107):
wraps any single-objective designer factory, scalarizing the observed
metrics before they reach it.
"""

from __future__ import annotations

from typing import Callable, Optional, Sequence

import numpy as np

from vizier_amd import pyvizier as vz
from vizier_amd._src.algorithms.core.abstractions import (
    ActiveTrials,
    CompletedTrials,
    Designer,
)
from vizier_amd._src.algorithms.designers.scalarization import (
    Scalarization,
)

_SCALARIZED_METRIC = 'scalarized'


class ScalarizingDesigner(Designer):
  """Presents a scalarized single-objective view to the inner designer."""

  def __init__(self, problem: vz.ProblemStatement,
               designer_factory: Callable[[vz.ProblemStatement], Designer],
               scalarization: Scalarization):
    self._problem = problem
    self._metrics = list(problem.metric_information)
    self._scalarization = scalarization
    self._scalarized_problem = vz.ProblemStatement(
        search_space=problem.search_space,
        metric_information=[vz.MetricInformation(
            name=_SCALARIZED_METRIC,
            goal=vz.ObjectiveMetricGoal.MAXIMIZE)],
        metadata=problem.metadata)
    self._designer = designer_factory(self._scalarized_problem)

  def update(self, completed: CompletedTrials, all_active: ActiveTrials
             ) -> None:
    converted = []
    for t in completed.trials:
      clone = vz.Trial(t.parameters, id=t.id, metadata=t.metadata)
      if t.final_measurement is not None and not t.infeasible:
        ys = []
        ok = True
        for mi in self._metrics:
          m = t.final_measurement.metrics.get(mi.name)
          if m is None:
            ok = False
            break
          ys.append(m.value if mi.goal.is_maximize else -m.value)
        if ok:
          value = float(self._scalarization(np.asarray(ys)))
          clone.complete(vz.Measurement(
              metrics={_SCALARIZED_METRIC: value}))
        else:
          clone.complete(vz.Measurement(), infeasibility_reason='missing')
      else:
        clone.complete(vz.Measurement(),
                       infeasibility_reason=t.infeasibility_reason
                       or 'infeasible')
      converted.append(clone)
    self._designer.update(CompletedTrials(converted), all_active)

  def suggest(self, count: Optional[int] = None
              ) -> Sequence[vz.TrialSuggestion]:
    return self._designer.suggest(count)
