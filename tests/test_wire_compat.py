"""Wire-compatibility check against the reference .proto files.

Parses the reference's proto sources (text only — no protoc in the
image) and verifies that every message field in our runtime-built
descriptors matches the reference's field NAME, NUMBER, LABEL and type
kind. Field numbers + wire types are the wire contract; this test is
the automated proof that the schemas agree.
"""

import os
import re

import pytest
from google.protobuf import descriptor_pool

REFERENCE_DIR = '/root/reference/vizier/_src/service'

_SCALARS = {
    'double', 'float', 'int32', 'int64', 'uint32', 'uint64', 'sint32',
    'sint64', 'fixed32', 'fixed64', 'sfixed32', 'sfixed64', 'bool',
    'string', 'bytes',
}


def _strip_comments(text: str) -> str:
  # Line comments only: '/*' appears inside http-option path strings
  # (e.g. "owners/*/studies/*"), and the reference uses '//' throughout.
  return re.sub(r'//[^\n]*', '', text)


def _parse_block(text: str, start: int):
  """Returns (body, end) for the brace block starting at text[start]=='{'."""
  depth = 0
  for i in range(start, len(text)):
    if text[i] == '{':
      depth += 1
    elif text[i] == '}':
      depth -= 1
      if depth == 0:
        return text[start + 1:i], i + 1
  raise ValueError('Unbalanced braces')


_FIELD_RE = re.compile(
    r'(?P<label>optional\s+|repeated\s+)?'
    r'(?P<type>[\w.]+)\s+(?P<name>\w+)\s*=\s*(?P<number>\d+)')


def _parse_messages(text: str, prefix: str, out: dict):
  pos = 0
  while True:
    m = re.search(r'\b(message|enum)\s+(\w+)\s*{', text[pos:])
    if not m:
      break
    kind, name = m.group(1), m.group(2)
    body, endpos = _parse_block(text, pos + m.end() - 1)
    full = f'{prefix}.{name}'
    if kind == 'message':
      fields = {}
      # Remove nested blocks before scanning fields at this level.
      flat = ''
      i = 0
      while i < len(body):
        nested = re.search(r'\b(message|enum|oneof)\s+\w+\s*{', body[i:])
        if not nested:
          flat += body[i:]
          break
        flat += body[i:i + nested.start()]
        nested_body, nested_end = _parse_block(body,
                                               i + nested.end() - 1)
        if nested.group(1) == 'oneof':
          flat += nested_body  # oneof fields live at message level
        i = i + nested.end() - 1 + (nested_end - (i + nested.end() - 1))
      for fm in _FIELD_RE.finditer(flat):
        if fm.group('type') in ('option', 'reserved'):
          continue
        fields[fm.group('name')] = {
            'number': int(fm.group('number')),
            'type': fm.group('type'),
            'repeated': (fm.group('label') or '').strip() == 'repeated',
        }
      out[full] = fields
      _parse_messages(body, full, out)
    pos = pos + m.start() + 1 if kind == 'enum' else endpos
    if kind == 'enum':
      pos = endpos


def _reference_schema() -> dict:
  out = {}
  for fname in ('key_value.proto', 'study.proto', 'vizier_oss.proto',
                'vizier_service.proto', 'pythia_service.proto'):
    path = os.path.join(REFERENCE_DIR, fname)
    text = _strip_comments(open(path).read())
    _parse_messages(text, 'vizier', out)
  return out


@pytest.mark.skipif(not os.path.isdir(REFERENCE_DIR),
                    reason='reference protos not available')
def test_every_reference_field_matches():
  import vizier_amd._src.service.proto.schema  # registers descriptors
  pool = descriptor_pool.Default()
  reference = _reference_schema()
  assert len(reference) > 30, 'parser failed to find reference messages'
  mismatches = []
  for full_name, fields in reference.items():
    try:
      desc = pool.FindMessageTypeByName(full_name)
    except KeyError:
      mismatches.append(f'missing message {full_name}')
      continue
    ours = {f.name: f for f in desc.fields}
    for fname, ref in fields.items():
      if fname not in ours:
        mismatches.append(f'{full_name}.{fname}: missing')
        continue
      f = ours[fname]
      if f.number != ref['number']:
        mismatches.append(
            f'{full_name}.{fname}: number {f.number} != {ref["number"]}')
      if f.is_repeated != ref['repeated']:
        mismatches.append(f'{full_name}.{fname}: repeated mismatch')
      ref_type = ref['type']
      if ref_type in _SCALARS:
        from google.protobuf import descriptor as _d
        expected = getattr(_d.FieldDescriptor,
                           'TYPE_' + ref_type.upper())
        if f.type != expected:
          mismatches.append(
              f'{full_name}.{fname}: scalar type {f.type} != {ref_type}')
      else:
        # Message or enum: compare the last name component.
        target = (f.message_type.name if f.message_type is not None
                  else f.enum_type.name if f.enum_type is not None
                  else None)
        if target != ref_type.split('.')[-1]:
          mismatches.append(
              f'{full_name}.{fname}: target {target} != {ref_type}')
  assert not mismatches, '\n'.join(mismatches)


@pytest.mark.skipif(not os.path.isdir(REFERENCE_DIR),
                    reason='reference protos not available')
def test_grpc_method_paths_match_reference_services():
  """Service + method names from the reference protos match our stubs."""
  from vizier_amd._src.service import service_stubs
  text = _strip_comments(
      open(os.path.join(REFERENCE_DIR, 'vizier_service.proto')).read())
  methods = re.findall(r'rpc\s+(\w+)\s*\(', text)
  assert set(methods) == set(service_stubs.VIZIER_METHODS)
  text = _strip_comments(
      open(os.path.join(REFERENCE_DIR, 'pythia_service.proto')).read())
  methods = re.findall(r'rpc\s+(\w+)\s*\(', text)
  assert set(methods) == set(service_stubs.PYTHIA_METHODS)
