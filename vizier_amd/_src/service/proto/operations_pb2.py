"""Accessors for the wire-compatible google.longrunning subset."""

from vizier_amd._src.service.proto.schema import get_message_class

Operation = get_message_class('google.longrunning.Operation')
GetOperationRequest = get_message_class('google.longrunning.GetOperationRequest')
Status = get_message_class('google.rpc.Status')
