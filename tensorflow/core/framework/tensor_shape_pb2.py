"""Import-path parity shim: re-exports wire-layer classes under the
reference-generated module path (reference setup.py:42-49 generates these
with protoc; here they are thin re-export modules over
min_tfs_client_amd.wire)."""
from min_tfs_client_amd.wire import messages as _m
TensorShapeProto = _m.TensorShapeProto
