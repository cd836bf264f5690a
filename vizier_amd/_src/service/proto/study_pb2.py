"""Accessors for study.proto / key_value.proto messages."""

from vizier_amd._src.service.proto.schema import get_message_class

KeyValue = get_message_class('vizier.KeyValue')
Study = get_message_class('vizier.Study')
Trial = get_message_class('vizier.Trial')
StudySpec = get_message_class('vizier.StudySpec')
Measurement = get_message_class('vizier.Measurement')
