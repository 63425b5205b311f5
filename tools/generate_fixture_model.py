#!/usr/bin/env python3
"""Generates the integration-test fixture model directory (the analogue of
reference tests/integration/fixtures/generate_tensorflow_model.py:12-57,
which builds a TF1 identity SavedModel): an identity servable version dir
with a warmup file, in this framework's repository layout."""
import os
import sys

import numpy as np

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

from min_tfs_client_amd.repository import write_warmup_file  # noqa: E402
from min_tfs_client_amd.tensors import ndarray_to_tensor_proto  # noqa: E402
from min_tfs_client_amd.wire import messages as pb  # noqa: E402


def main(base="tests/integration/fixtures/default"):
    vdir = os.path.join(base, "00000001")
    os.makedirs(vdir, exist_ok=True)
    open(os.path.join(vdir, "identity"), "w").close()
    req = pb.PredictRequest()
    req.model_spec.name = "default"
    req.inputs["string_input"].CopyFrom(
        ndarray_to_tensor_proto(np.array(["warmup"])))
    req.inputs["float_input"].CopyFrom(
        ndarray_to_tensor_proto(np.zeros((1, 100), np.float32)))
    req.inputs["int_input"].CopyFrom(
        ndarray_to_tensor_proto(np.array([1], np.int64)))
    write_warmup_file(vdir, [req])
    print(f"fixture written to {vdir}")


if __name__ == "__main__":
    main(*sys.argv[1:])
