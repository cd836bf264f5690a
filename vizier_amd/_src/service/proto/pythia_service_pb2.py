"""Accessors for pythia_service.proto messages."""

from vizier_amd._src.service.proto.schema import get_message_class

TrialSuggestion = get_message_class('vizier.TrialSuggestion')
ProblemStatement = get_message_class('vizier.ProblemStatement')
StudyDescriptor = get_message_class('vizier.StudyDescriptor')
SuggestRequest = get_message_class('vizier.SuggestRequest')
SuggestDecision = get_message_class('vizier.SuggestDecision')
EarlyStopRequest = get_message_class('vizier.EarlyStopRequest')
EarlyStopDecision = get_message_class('vizier.EarlyStopDecision')
EarlyStopDecisions = get_message_class('vizier.EarlyStopDecisions')
