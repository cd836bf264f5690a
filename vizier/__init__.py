"""Drop-in import alias: `vizier` -> `vizier_amd`.

Lets code written against the reference's import paths
(`from vizier import pyvizier as vz`, `from vizier.service import
clients`, ...) run unchanged on this MI355X-native implementation.
"""

import sys as _sys

import vizier_amd as _impl
import vizier_amd.pyvizier as _pyvizier
import vizier_amd.pythia as _pythia
import vizier_amd.algorithms as _algorithms
import vizier_amd.benchmarks as _benchmarks
import vizier_amd.service as _service
import vizier_amd.client as _client
import vizier_amd.converters as _converters
import vizier_amd.interfaces as _interfaces
import vizier_amd.utils as _utils
import vizier_amd.testing as _testing
import vizier_amd.raytune as _raytune
import vizier_amd.pyglove as _pyglove
import vizier_amd.benchmarks.experimenters as _bench_experimenters
import vizier_amd.benchmarks.analyzers as _bench_analyzers
import vizier_amd.service.clients as _service_clients
import vizier_amd.service.servers as _service_servers
import vizier_amd.service.pyvizier as _service_pyvizier

__version__ = _impl.__version__

_sys.modules[__name__ + '.pyvizier'] = _pyvizier
_sys.modules[__name__ + '.pythia'] = _pythia
_sys.modules[__name__ + '.algorithms'] = _algorithms
_sys.modules[__name__ + '.benchmarks'] = _benchmarks
_sys.modules[__name__ + '.service'] = _service
_sys.modules[__name__ + '.client'] = _client
_sys.modules[__name__ + '.converters'] = _converters
_sys.modules[__name__ + '.interfaces'] = _interfaces
_sys.modules[__name__ + '.utils'] = _utils
_sys.modules[__name__ + '.testing'] = _testing
_sys.modules[__name__ + '.raytune'] = _raytune
_sys.modules[__name__ + '.pyglove'] = _pyglove
_sys.modules[__name__ + '.benchmarks.experimenters'] = _bench_experimenters
_sys.modules[__name__ + '.benchmarks.analyzers'] = _bench_analyzers
_sys.modules[__name__ + '.service.clients'] = _service_clients
_sys.modules[__name__ + '.service.servers'] = _service_servers
_sys.modules[__name__ + '.service.pyvizier'] = _service_pyvizier

pyvizier = _pyvizier
pythia = _pythia
algorithms = _algorithms
benchmarks = _benchmarks
service = _service
client = _client
raytune = _raytune
pyglove = _pyglove
converters = _converters
interfaces = _interfaces
utils = _utils
testing = _testing
