"""Import-path parity shim: re-exports wire-layer classes under the
reference-generated module path (TF generates
tensorflow/core/profiler/profiler_service_pb2.py; here a thin re-export
over min_tfs_client_amd.wire)."""
from min_tfs_client_amd.wire import messages as _m
ProfileOptions = _m.ProfileOptions
ToolRequestOptions = _m.ToolRequestOptions
ProfileRequest = _m.ProfileRequest
ProfileToolData = _m.ProfileToolData
ProfileResponse = _m.ProfileResponse
MonitorRequest = _m.MonitorRequest
MonitorResponse = _m.MonitorResponse
