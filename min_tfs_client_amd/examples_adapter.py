"""Classify / Regress over tf.Example inputs.

The reference's Classify/Regress rpcs take ``tensorflow.serving.Input``
(example_list of tf.Example protos — input.proto:15-82) and return
Class/Regression results (classification.proto:11-48, regression.proto:11-37).
This module supplies:

* feature extraction: Example protos -> {feature_name: ndarray} batches
  (bytes_list / float_list / int64_list arms),
* ``ClassificationAdapter`` / ``RegressionAdapter``: wrap any tensor
  servable so it serves the Classify / Regress rpcs (scores from a named
  output; optional label vocabulary),
* ``examples_input(...)`` client helper building an Input proto from
  python dicts.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import numpy as np

from .server import Servable
from .wire import messages as pb


# ---------------------------------------------------------------------------
# client-side helper
# ---------------------------------------------------------------------------

def make_example(features: Dict) -> "pb.Example":
    ex = pb.Example()
    for name, value in features.items():
        f = ex.features.feature[name]
        if isinstance(value, (bytes, str)):
            v = value.encode() if isinstance(value, str) else value
            f.bytes_list.value.append(v)
        elif isinstance(value, (list, tuple, np.ndarray)):
            arr = np.asarray(value)
            if arr.dtype.kind in "iu":
                f.int64_list.value.extend(int(x) for x in arr.ravel())
            elif arr.dtype.kind == "f":
                f.float_list.value.extend(float(x) for x in arr.ravel())
            else:
                f.bytes_list.value.extend(
                    x.encode() if isinstance(x, str) else bytes(x)
                    for x in arr.ravel())
        elif isinstance(value, (int, np.integer)):
            f.int64_list.value.append(int(value))
        elif isinstance(value, (float, np.floating)):
            f.float_list.value.append(float(value))
        else:
            raise TypeError(f"unsupported feature value for {name!r}")
    return ex


def examples_input(rows: Sequence[Dict]) -> "pb.Input":
    """[{feature: value}] -> Input{example_list}."""
    inp = pb.Input()
    for row in rows:
        inp.example_list.examples.add().CopyFrom(make_example(row))
    return inp


# ---------------------------------------------------------------------------
# server-side feature extraction
# ---------------------------------------------------------------------------

def examples_to_feature_arrays(input_proto) -> Dict[str, np.ndarray]:
    """Input{example_list|example_list_with_context} -> batched arrays.
    Context features (ExampleListWithContext, input.proto:29-66) are merged
    into every example, like TF's input processing."""
    kind = input_proto.WhichOneof("kind")
    if kind == "example_list":
        examples = list(input_proto.example_list.examples)
        context = None
    elif kind == "example_list_with_context":
        examples = list(input_proto.example_list_with_context.examples)
        context = input_proto.example_list_with_context.context
    else:
        raise ValueError("Input is empty (expected example_list)")
    if not examples:
        raise ValueError("Input batch is empty")

    names = set()
    for ex in examples:
        names.update(ex.features.feature.keys())
    if context is not None:
        names.update(context.features.feature.keys())

    out: Dict[str, np.ndarray] = {}
    for name in sorted(names):
        cols = []
        for ex in examples:
            feat = None
            if name in ex.features.feature:
                feat = ex.features.feature[name]
            elif context is not None and name in context.features.feature:
                feat = context.features.feature[name]
            if feat is None:
                raise ValueError(f"feature {name!r} missing from an example")
            arm = feat.WhichOneof("kind")
            if arm == "int64_list":
                cols.append(np.asarray(feat.int64_list.value,
                                       dtype=np.int64))
            elif arm == "float_list":
                cols.append(np.asarray(feat.float_list.value,
                                       dtype=np.float32))
            elif arm == "bytes_list":
                cols.append(np.asarray(list(feat.bytes_list.value),
                                       dtype=object))
            else:
                raise ValueError(f"feature {name!r} has no value")
        width = {len(c) for c in cols}
        if len(width) != 1:
            raise ValueError(f"feature {name!r} has ragged widths {width}")
        out[name] = np.stack(cols)
    return out


# ---------------------------------------------------------------------------
# adapters
# ---------------------------------------------------------------------------

class ClassificationAdapter(Servable):
    """Makes a tensor servable classify tf.Examples: features are
    extracted, the inner servable runs, and ``scores_output`` (shape
    [batch, n_classes]) becomes Classifications with optional labels."""

    def __init__(self, inner: Servable, scores_output: str = "scores",
                 labels: Optional[List[str]] = None):
        super().__init__(inner.fn, inner.signature, inner.signature_name)
        self.inner = inner
        self.scores_output = scores_output
        self.labels = labels

    def classify(self, input_proto) -> "pb.ClassificationResult":
        features = examples_to_feature_arrays(input_proto)
        outputs = self.inner(features)
        if self.scores_output not in outputs:
            raise ValueError(
                f"Expected classification scores output "
                f"{self.scores_output!r}; servable returned "
                f"{sorted(outputs)}")
        scores = np.asarray(outputs[self.scores_output])
        if scores.ndim == 1:
            scores = scores[:, None]
        result = pb.ClassificationResult()
        for row in scores:
            cls = result.classifications.add()
            for j, s in enumerate(row):
                c = cls.classes.add()
                c.label = (self.labels[j] if self.labels
                           and j < len(self.labels) else str(j))
                c.score = float(s)
        return result


class RegressionAdapter(Servable):
    """Regress over tf.Examples: ``value_output`` (shape [batch] or
    [batch,1]) becomes Regression values."""

    def __init__(self, inner: Servable, value_output: str = "value"):
        super().__init__(inner.fn, inner.signature, inner.signature_name)
        self.inner = inner
        self.value_output = value_output

    def regress(self, input_proto) -> "pb.RegressionResult":
        features = examples_to_feature_arrays(input_proto)
        outputs = self.inner(features)
        if self.value_output not in outputs:
            raise ValueError(
                f"Expected regression output {self.value_output!r}; "
                f"servable returned {sorted(outputs)}")
        values = np.asarray(outputs[self.value_output]).reshape(-1)
        result = pb.RegressionResult()
        for v in values:
            result.regressions.add().value = float(v)
        return result
