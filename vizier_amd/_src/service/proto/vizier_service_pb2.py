"""Accessors for vizier_service.proto messages."""

from vizier_amd._src.service.proto.schema import get_message_class

GetStudyRequest = get_message_class('vizier.GetStudyRequest')
CreateStudyRequest = get_message_class('vizier.CreateStudyRequest')
ListStudiesRequest = get_message_class('vizier.ListStudiesRequest')
ListStudiesResponse = get_message_class('vizier.ListStudiesResponse')
DeleteStudyRequest = get_message_class('vizier.DeleteStudyRequest')
SetStudyStateRequest = get_message_class('vizier.SetStudyStateRequest')
SuggestTrialsRequest = get_message_class('vizier.SuggestTrialsRequest')
SuggestTrialsResponse = get_message_class('vizier.SuggestTrialsResponse')
CreateTrialRequest = get_message_class('vizier.CreateTrialRequest')
GetTrialRequest = get_message_class('vizier.GetTrialRequest')
ListTrialsRequest = get_message_class('vizier.ListTrialsRequest')
ListTrialsResponse = get_message_class('vizier.ListTrialsResponse')
AddTrialMeasurementRequest = get_message_class(
    'vizier.AddTrialMeasurementRequest')
CompleteTrialRequest = get_message_class('vizier.CompleteTrialRequest')
DeleteTrialRequest = get_message_class('vizier.DeleteTrialRequest')
CheckTrialEarlyStoppingStateRequest = get_message_class(
    'vizier.CheckTrialEarlyStoppingStateRequest')
CheckTrialEarlyStoppingStateResponse = get_message_class(
    'vizier.CheckTrialEarlyStoppingStateResponse')
StopTrialRequest = get_message_class('vizier.StopTrialRequest')
ListOptimalTrialsRequest = get_message_class('vizier.ListOptimalTrialsRequest')
ListOptimalTrialsResponse = get_message_class(
    'vizier.ListOptimalTrialsResponse')
UnitMetadataUpdate = get_message_class('vizier.UnitMetadataUpdate')
UpdateMetadataRequest = get_message_class('vizier.UpdateMetadataRequest')
UpdateMetadataResponse = get_message_class('vizier.UpdateMetadataResponse')
