"""TunerPolicy: a Pythia policy backed by a pg.DNAGenerator.

Parity with vizier/_src/pyglove/pythia.py (TunerPolicy :30,
create_policy :180): newly-completed trials feed the pyglove algorithm
(reward = final measurement of the metrics-to-optimize), suggestions
come from `algorithm.propose()` converted to trials, DNA metadata
changes are persisted back into trial metadata, and the optional
pg.tuning.EarlyStoppingPolicy drives early_stop decisions.
"""

from __future__ import annotations

from typing import Optional, Sequence

import pyglove as pg

from vizier_amd import pyvizier as vz
from vizier_amd._src.pythia import policy as pythia_policy
from vizier_amd._src.pyglove import constants
from vizier_amd._src.pyglove import converters as cv
from vizier_amd._src.pyglove import core


class TunerPolicy(pythia_policy.Policy):
  """Adapts a pg.DNAGenerator to the Pythia Policy protocol."""

  def __init__(self, supporter, converter: cv.VizierConverter,
               algorithm, early_stopping_policy=None):
    self.supporter = supporter
    self._converter = converter
    self.algorithm = algorithm
    self.early_stopping_policy = early_stopping_policy
    self._incorporated_ids = set()

  @property
  def _metric_names(self) -> Sequence[str]:
    return self._converter.metrics_to_optimize

  def _newly_completed(self, max_trial_id: int):
    trials = self.supporter.GetTrials(
        max_trial_id=max_trial_id,
        status_matches=vz.TrialStatus.COMPLETED)
    return [t for t in trials if t.id not in self._incorporated_ids]

  def suggest(self, request: pythia_policy.SuggestRequest
              ) -> pythia_policy.SuggestDecision:
    metadata_updates = vz.MetadataDelta()
    for trial in self._newly_completed(request.max_trial_id):
      tuner_trial = core.VizierTrial(self._converter, trial)
      reward = tuner_trial.get_reward_for_feedback(self._metric_names)
      self._incorporated_ids.add(trial.id)
      if reward is None:
        continue
      before = dict(getattr(tuner_trial.dna, 'metadata', {}) or {})
      self.algorithm.feedback(tuner_trial.dna, reward)
      after = dict(getattr(tuner_trial.dna, 'metadata', {}) or {})
      if before != after:
        metadata_updates.assign(
            namespace=constants.METADATA_NAMESPACE,
            key=constants.TRIAL_METADATA_KEY_DNA_METADATA,
            value=pg.to_json_str(after), trial=trial)

    new_trials = []
    for _ in range(request.count or 1):
      try:
        dna = self.algorithm.propose()
      except StopIteration:
        break
      if getattr(dna, 'spec', None) is None and hasattr(dna, 'use_spec'):
        dna.use_spec(self._converter.dna_spec)
      trial = self._converter.to_trial(dna, fallback='return_dummy')
      new_trials.append(vz.TrialSuggestion(
          trial.parameters, metadata=trial.metadata))
    return pythia_policy.SuggestDecision(new_trials, metadata_updates)

  def early_stop(self, request: pythia_policy.EarlyStopRequest
                 ) -> pythia_policy.EarlyStopDecisions:
    decisions = pythia_policy.EarlyStopDecisions()
    if self.early_stopping_policy is None:
      return decisions
    for trial in self._newly_completed(request.max_trial_id):
      self.early_stopping_policy.should_stop_early(
          core.VizierTrial(self._converter, trial))
    active = self.supporter.GetTrials(
        status_matches=vz.TrialStatus.ACTIVE)
    for trial in active:
      should_stop = self.early_stopping_policy.should_stop_early(
          core.VizierTrial(self._converter, trial))
      decisions.decisions.append(pythia_policy.EarlyStopDecision(
          trial.id, should_stop=should_stop,
          reason='Pyglove stopping policy stopped the trial.'))
    return decisions


def create_policy(supporter, problem_statement: vz.ProblemStatement,
                  algorithm,
                  early_stopping_policy=None,
                  prior_trials: Optional[Sequence[vz.Trial]] = None
                  ) -> TunerPolicy:
  """Builds a TunerPolicy from a problem (pythia.py:180-202)."""
  converter = cv.VizierConverter.from_problem(problem_statement)
  algorithm.setup(converter.dna_spec)
  if prior_trials:
    def history():
      for trial in prior_trials:
        tuner_trial = core.VizierTrial(converter, trial)
        yield (tuner_trial.dna, tuner_trial.get_reward_for_feedback(
            converter.metrics_to_optimize))
    algorithm.recover(history())
  return TunerPolicy(supporter, converter, algorithm,
                     early_stopping_policy)
