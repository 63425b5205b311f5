#!/usr/bin/env python3
"""The secondary self-referential baseline from BASELINE.md: the reference
client's pure-python per-element encode of a 32x3x224x224 fp32 tensor
(reference tensors.py:23 — ``proto_field.extend([v.item() for v in
values])`` over 4.8M elements) versus this framework's paths, on the same
host.
"""
import json
import os
import statistics
import sys
import time

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import numpy as np  # noqa: E402
import torch  # noqa: E402

from min_tfs_client_amd import _native as native  # noqa: E402
from min_tfs_client_amd.tensors import ndarray_to_tensor_proto  # noqa: E402
from min_tfs_client_amd.wire import messages as pb  # noqa: E402


def timeit(fn, reps=5, warmup=1):
    for _ in range(warmup):
        fn()
    ts = []
    for _ in range(reps):
        t0 = time.perf_counter()
        fn()
        ts.append(time.perf_counter() - t0)
    return statistics.median(ts) * 1e3


def reference_encode(arr: np.ndarray) -> bytes:
    """Byte-for-byte what the reference client does (tensors.py:17-35):
    a python loop appending each element to the typed float_val field."""
    proto = pb.TensorProto(
        dtype=1,
        tensor_shape=pb.TensorShapeProto(
            dim=[pb.TensorShapeProto.Dim(size=d) for d in arr.shape]))
    proto.float_val.extend([v.item() for v in arr.ravel()])
    req = pb.PredictRequest()
    req.model_spec.name = "m"
    req.inputs["x"].CopyFrom(proto)
    return req.SerializeToString()


def main():
    arr = np.random.rand(32, 3, 224, 224).astype(np.float32)
    t = torch.from_numpy(arr)
    results = {"shape": list(arr.shape), "elements": int(arr.size)}

    # 1. reference per-element loop (one reverent rep: it is slow)
    results["reference_per_element_ms"] = timeit(
        lambda: reference_encode(arr), reps=1, warmup=0)

    # 2. our python codec, typed fields (vectorized tolist)
    results["python_typed_ms"] = timeit(
        lambda: ndarray_to_tensor_proto(arr, use_tensor_content=False),
        reps=3)

    # 3. our python codec, tensor_content
    results["python_tensor_content_ms"] = timeit(
        lambda: ndarray_to_tensor_proto(arr), reps=5)

    # 4. C++ codec, full PredictRequest bytes (CPU tensor)
    results["native_serialize_cpu_ms"] = timeit(
        lambda: native.serialize_predict_request("m", -1, "", ["x"], [t], 0),
        reps=10)

    # 5. C++ codec from the GPU (HIP pack + staged D2H)
    if torch.cuda.is_available():
        g = t.cuda()
        results["native_serialize_gpu_ms"] = timeit(
            lambda: native.serialize_predict_request(
                "m", -1, "", ["x"], [g], 0), reps=10)

    ref = results["reference_per_element_ms"]
    results["speedup_native_cpu_vs_reference"] = round(
        ref / results["native_serialize_cpu_ms"], 1)
    if "native_serialize_gpu_ms" in results:
        results["speedup_native_gpu_vs_reference"] = round(
            ref / results["native_serialize_gpu_ms"], 1)
    print(json.dumps(results, indent=1))


if __name__ == "__main__":
    main()
