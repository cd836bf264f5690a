"""OSS Vizier pyglove backend: `vizier_amd.pyglove.init()` + backend.

Parity with vizier/_src/pyglove/oss_vizier.py (_OSSVizierTuner :174,
init :264, OSSVizierBackend :290): wires the pyglove tuning loop into
THIS service — in-process (NO_ENDPOINT) by default, or against a
remote endpoint — and registers the backend with `pg.tuning` under the
'oss_vizier' name so `pg.sample(..., backend='oss_vizier')` works.

In-process mode hosts the "Pythia service" by installing a
PyGlovePolicyFactory into the local VizierServicer's pythia servicer:
studies whose key is in the policy cache run the cached TunerPolicy
(the pyglove algorithm); everything else falls through to the default
algorithm registry.
"""

from __future__ import annotations

import threading
from typing import Optional, Union

import pyglove as pg

from vizier_amd import pyvizier as vz
from vizier_amd._src.pyglove import algorithms
from vizier_amd._src.pyglove import backend
from vizier_amd._src.pyglove import client
from vizier_amd._src.pyglove import constants
from vizier_amd._src.pyglove import converters
from vizier_amd._src.service import clients as service_clients
from vizier_amd._src.service import policy_factory as default_factory_lib
from vizier_amd._src.service import pythia_service
from vizier_amd._src.service import resources
from vizier_amd._src.service import service_policy_supporter
from vizier_amd._src.service import vizier_client

BuiltinAlgorithm = algorithms.BuiltinAlgorithm
ExpandedStudyName = client.ExpandedStudyName
PolicyCache = client.PolicyCache
StudyKey = client.StudyKey


class PyGlovePolicyFactory:
  """Routes cached pyglove studies to their TunerPolicy (:57-85)."""

  def __init__(self, policy_cache: PolicyCache):
    self._policy_cache = policy_cache
    self._default = default_factory_lib.DefaultPolicyFactory()

  def __call__(self, problem_statement, algorithm, policy_supporter,
               study_name):
    study_resource = resources.StudyResource.from_name(study_name)
    key = StudyKey(study_resource.owner_id,
                   ExpandedStudyName(study_resource.study_id))
    if key in self._policy_cache:
      return self._policy_cache[key]
    return self._default(problem_statement, algorithm, policy_supporter,
                         study_name)


class _VizierServices:
  """Process-wide service hub (oss_vizier.py:88)."""

  def __init__(self):
    self._vizier_service = None
    self._pythia_started = False
    self._live_tuners = set()

  def reset_for_testing(self) -> None:
    self.__init__()
    vizier_client._create_local_vizier_servicer.cache_clear()

  def use_vizier_service(self, endpoint: Optional[str]) -> None:
    if endpoint:
      service_clients_env = vizier_client.environment_variables
      service_clients_env.server_endpoint = endpoint
    if self._vizier_service is None:
      self._vizier_service = \
          vizier_client.create_vizier_servicer_or_stub()

  @property
  def vizier_service(self):
    assert self._vizier_service is not None, \
        'call use_vizier_service first'
    return self._vizier_service

  def start_pythia_service(self, policy_cache: PolicyCache) -> None:
    """Installs the pyglove-aware policy factory (in-process mode)."""
    if self._pythia_started:
      return
    servicer = pythia_service.PythiaServicer(
        vizier_service=self._vizier_service,
        policy_factory=PyGlovePolicyFactory(policy_cache))
    if hasattr(self._vizier_service, 'default_pythia_service'):
      self._vizier_service.default_pythia_service = servicer
    self._pythia_started = True

  def register_tuner(self, tuner_id: str) -> None:
    self._live_tuners.add(tuner_id)

  def tuner_alive(self, tuner_id: str) -> bool:
    return tuner_id in self._live_tuners

  def drop_tuner(self, tuner_id: str) -> None:
    self._live_tuners.discard(tuner_id)


_services = _VizierServices()


class _OSSVizierTuner(client.VizierTuner):
  """In-process OSS tuner (oss_vizier.py:174)."""

  def get_tuner_id(self, algorithm) -> str:
    del algorithm
    tuner_id = f'{threading.get_ident()}@local'
    _services.register_tuner(tuner_id)
    return tuner_id

  def _start_pythia_service(self, policy_cache: PolicyCache) -> None:
    _services.start_pythia_service(policy_cache)

  def load_prior_study(self, resource_name: str):
    return service_clients.Study.from_resource_name(resource_name)

  @classmethod
  def load_study(cls, owner: str, name: ExpandedStudyName):
    return service_clients.Study.from_owner_and_id(owner, str(name))

  def create_study(self, problem: vz.ProblemStatement, converter,
                   owner: str, name: str, algorithm,
                   stopping_policy=None):
    study_config = vz.StudyConfig(
        search_space=problem.search_space,
        metric_information=problem.metric_information,
        metadata=problem.metadata)
    if getattr(converter, 'vizier_conversion_error', None):
      study_config.observation_noise = vz.ObservationNoise.HIGH
    if isinstance(algorithm, algorithms.BuiltinAlgorithm):
      study_config.algorithm = algorithm.name
    else:
      study_config.algorithm = 'EXTERNAL_PYTHIA_SERVICE'
    return service_clients.Study.from_study_config(
        study_config, owner=owner, study_id=name)

  def get_group_id(self, group_id: Union[None, int, str] = None) -> str:
    if group_id is None:
      return f'{threading.get_ident()}@local'
    if isinstance(group_id, int):
      return f'group:{group_id}'
    return group_id

  def ping_tuner(self, tuner_id: str) -> bool:
    return _services.tuner_alive(tuner_id)

  def pythia_supporter(self, study):
    return service_policy_supporter.ServicePolicySupporter(
        study.resource_name, _services.vizier_service)

  def use_pythia_for_study(self, study) -> None:
    # In-process: the local servicer's pythia already consults the
    # policy cache; nothing endpoint-specific to record.
    pass


def init(study_prefix: Optional[str] = None,
         vizier_endpoint: Optional[str] = None,
         pythia_port: Optional[int] = None) -> None:
  """Initializes the OSS Vizier pyglove backend (oss_vizier.py:264)."""
  del pythia_port  # In-process Pythia needs no port.
  _services.use_vizier_service(vizier_endpoint)
  backend.VizierBackend.use_study_prefix(study_prefix)
  if hasattr(pg, 'tuning') and hasattr(pg.tuning, 'set_default_backend'):
    pg.tuning.set_default_backend('oss_vizier')


class OSSVizierBackend(backend.VizierBackend):
  """PyGlove backend that uses OSS Vizier (oss_vizier.py:290)."""

  tuner_cls = _OSSVizierTuner


if hasattr(pg, 'tuning') and hasattr(pg.tuning, 'add_backend'):
  try:
    pg.tuning.add_backend('oss_vizier')(OSSVizierBackend)
  except TypeError:
    # Some pyglove versions use add_backend as a plain registrar.
    pg.tuning.add_backend('oss_vizier', OSSVizierBackend)
