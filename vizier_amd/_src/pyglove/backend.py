"""Cross-platform pyglove tuning backend with chief election.

Parity with vizier/_src/pyglove/backend.py (VizierBackend :70): each
worker connects to (or creates) the shared study; the CHIEF — elected
through the study-metadata PRIMARY_TUNER_ID key — hosts the Pythia
policy for non-builtin algorithms, and any worker that finds the
recorded chief unreachable (ping fails) registers itself as the new
primary. Workers pull suggestions through the regular clients API
using their group's client_id, so batching/deduplication follows the
service's REQUESTED-pool semantics.
"""

from __future__ import annotations

import getpass
from typing import Any, Dict, Optional, Sequence, Type, Union

import pyglove as pg

from vizier_amd import pyvizier as vz
from vizier_amd._src.pyglove import algorithms
from vizier_amd._src.pyglove import client
from vizier_amd._src.pyglove import constants
from vizier_amd._src.pyglove import converters
from vizier_amd._src.pyglove import core
from vizier_amd._src.pyglove import tuner_policy

TunerPolicy = tuner_policy.TunerPolicy
BuiltinAlgorithm = algorithms.BuiltinAlgorithm
ExpandedStudyName = client.ExpandedStudyName
StudyKey = client.StudyKey

# Global policy cache shared with the Pythia servicer.
_global_policy_cache: Dict[StudyKey, TunerPolicy] = {}


class VizierBackend:
  """pg.tuning backend over the Vizier service (backend.py:70)."""

  default_owner: str = getpass.getuser()
  default_study_prefix: Optional[str] = None
  tuner_cls: Type[client.VizierTuner] = None

  def __init__(self, name: Optional[str],
               group: Union[None, int, str],
               dna_spec, algorithm,
               metrics_to_optimize: Sequence[str],
               early_stopping_policy=None,
               num_examples: Optional[int] = None,
               study_owner: Optional[str] = None,
               prior_study_ids: Optional[Sequence[str]] = None,
               add_prior_trials: bool = False,
               is_chief: Optional[bool] = None):
    self._tuner = self.tuner_cls()
    self._dna_spec = dna_spec
    self._algorithm = algorithm
    self._early_stopping_policy = early_stopping_policy
    self._group = group
    self._num_examples = num_examples
    self._prior_study_ids = tuple(prior_study_ids or ())
    self._add_prior_trials = add_prior_trials
    self._study_owner = study_owner or self.default_owner
    self._study_name = self._expand_name(name)
    self._converter = converters.VizierConverter.from_dna_spec(
        dna_spec, metrics_to_optimize)
    self._forced_chief = is_chief
    self._is_active = True

    is_chief_now = self._setup_study()
    self._suggestion_generator = self._create_suggestion_generator()
    if self._need_pythia_service and is_chief_now:
      self._start_pythia()

  # -- study setup + election ------------------------------------------------

  @property
  def _host_pythia_algorithm(self) -> bool:
    return not isinstance(self._algorithm, algorithms.PseudoAlgorithm)

  @property
  def _need_pythia_service(self) -> bool:
    return self._host_pythia_algorithm

  @property
  def _tuner_id(self) -> str:
    return self._tuner.get_tuner_id(self._algorithm)

  def _setup_study(self) -> bool:
    """Loads or creates the study; returns True when we are chief."""
    from vizier_amd._src.service import custom_errors
    try:
      self._study = self._tuner.load_study(self._study_owner,
                                           self._study_name)
      stored = self._get_stored_dna_spec()
      if stored is not None and pg.to_json(stored) != pg.to_json(
          self._dna_spec):
        raise ValueError(
            'The client-side search space is different from the one '
            'stored in the study. Use a different study name.')
      chief = self._get_chief_tuner_id()
      if self._forced_chief is not False and not self._tuner.ping_tuner(
          chief):
        # The recorded chief is gone: elect ourselves (races are
        # benign — all state lives in the study; see reference
        # backend.py:446-460).
        chief = self._register_self_as_primary()
      is_chief = chief == self._tuner_id
      if self._forced_chief is False and is_chief:
        raise ValueError(
            f'{self._tuner_id!r} runs as secondary but the study says '
            'it is primary.')
      return is_chief
    except (KeyError, LookupError, custom_errors.NotFoundError):
      if self._forced_chief is False:
        self._study = self._wait_for_study()
        return False
      problem = self._converter.problem_or_dummy
      problem.metadata.ns(constants.METADATA_NAMESPACE)[
          constants.STUDY_METADATA_KEY_TUNER_ID] = self._tuner_id
      self._study = self._tuner.create_study(
          problem, self._converter, self._study_owner, self._study_name,
          self._algorithm, self._early_stopping_policy)
      is_chief = self._tuner_id == self._get_chief_tuner_id()
      if is_chief and self._add_prior_trials:
        for trial in self._load_prior_trials():
          self._study.add_trial(trial)
      return is_chief

  def _wait_for_study(self):
    from vizier_amd._src.service import custom_errors
    import time
    while True:
      try:
        return self._tuner.load_study(self._study_owner,
                                      self._study_name)
      except (KeyError, LookupError, custom_errors.NotFoundError):
        time.sleep(1.0)

  def _get_stored_dna_spec(self):
    metadata = self._study.materialize_problem_statement().metadata.ns(
        constants.METADATA_NAMESPACE)
    blob = metadata.get(constants.STUDY_METADATA_KEY_DNA_SPEC, None)
    return None if blob is None else converters.restore_dna_spec(blob)

  def _get_chief_tuner_id(self) -> str:
    metadata = self._study.materialize_problem_statement().metadata.ns(
        constants.METADATA_NAMESPACE)
    try:
      return str(metadata[constants.STUDY_METADATA_KEY_TUNER_ID])
    except KeyError as e:
      raise RuntimeError(
          f'{constants.STUDY_METADATA_KEY_TUNER_ID} missing in study '
          f'{self._study.resource_name}.') from e

  def _register_self_as_primary(self) -> str:
    metadata = vz.Metadata()
    metadata.ns(constants.METADATA_NAMESPACE)[
        constants.STUDY_METADATA_KEY_TUNER_ID] = self._tuner_id
    self._study.update_metadata(metadata)
    self._tuner.use_pythia_for_study(self._study)
    return self._tuner_id

  # -- pythia hosting --------------------------------------------------------

  def _start_pythia(self) -> None:
    self._tuner.start_pythia_service(_global_policy_cache)
    self._algorithm.setup(self._dna_spec)
    prior_trials = ()
    if not self._add_prior_trials:
      prior_trials = self._load_prior_trials()
    if prior_trials:
      def history():
        for trial in prior_trials:
          tuner_trial = core.VizierTrial(self._converter, trial)
          yield (tuner_trial.dna, tuner_trial.get_reward_for_feedback(
              self._converter.metrics_to_optimize))
      self._algorithm.recover(history())
    policy = TunerPolicy(self._tuner.pythia_supporter(self._study),
                         self._converter, self._algorithm,
                         self._early_stopping_policy)
    key = StudyKey(self._study_owner, self._study_name)
    existing = _global_policy_cache.get(key)
    if existing is not None and existing.algorithm != policy.algorithm:
      raise ValueError(
          f'Different algorithms used for the same study {key!r}.')
    _global_policy_cache[key] = policy

  def _load_prior_trials(self):
    trials = []
    for prior in self._prior_study_ids:
      study = self._tuner.load_prior_study(prior)
      trials.extend(study.trials(vz.TrialFilter(
          status=vz.TrialStatus.COMPLETED)).get())
    return trials

  # -- sampling loop ---------------------------------------------------------

  def _create_suggestion_generator(self):
    while (self._num_examples is None or
           len(list(self._study.trials())) < self._num_examples):
      trials = self._study.suggest(
          count=1, client_id=self._tuner.get_group_id(self._group))
      if not trials:
        return
      for trial in trials:
        yield core.Feedback(trial, self._converter)

  def next(self) -> core.Feedback:
    try:
      return next(self._suggestion_generator)
    except StopIteration:
      self._is_active = False
      raise

  # -- class helpers ---------------------------------------------------------

  @classmethod
  def use_study_prefix(cls, study_prefix: Optional[str]) -> None:
    cls.default_study_prefix = study_prefix or ''

  @classmethod
  def _expand_name(cls, name: Optional[str]) -> str:
    components = []
    if cls.default_study_prefix:
      components.append(cls.default_study_prefix)
    if name:
      components.append(name)
    return ExpandedStudyName('.'.join(components))

  @classmethod
  def poll_result(cls, name: str,
                  study_owner: Optional[str] = None) -> core.Result:
    return core.Result.from_study(cls.tuner_cls.load_study(
        study_owner or cls.default_owner, cls._expand_name(name)))
