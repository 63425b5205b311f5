"""Import-path parity shim: re-exports wire-layer classes under the
reference-generated module path (reference setup.py:42-49 generates these
with protoc; here they are thin re-export modules over
min_tfs_client_amd.wire)."""
from min_tfs_client_amd.wire import messages as _m
BytesList = _m.BytesList
FloatList = _m.FloatList
Int64List = _m.Int64List
Feature = _m.Feature
Features = _m.Features
FeatureList = _m.FeatureList
FeatureLists = _m.FeatureLists
