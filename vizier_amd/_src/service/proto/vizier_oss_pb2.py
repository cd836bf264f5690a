"""Accessors for vizier_oss.proto messages."""

from vizier_amd._src.service.proto.schema import get_message_class

EarlyStoppingOperation = get_message_class('vizier.EarlyStoppingOperation')
