"""Serializer round-trips (parity with the reference's
test_serializer_from_definition / into_definition / dump-load tests)."""
import numpy as np
import pytest
import yaml
from sklearn.pipeline import FeatureUnion, Pipeline
from sklearn.preprocessing import MinMaxScaler

from gordo_amd import serializer
from gordo_amd.model.models import TorchAutoEncoder


def test_from_definition_simple_string():
    obj = serializer.from_definition("sklearn.preprocessing.MinMaxScaler")
    assert isinstance(obj, MinMaxScaler)


def test_from_definition_with_params():
    obj = serializer.from_definition(
        {"sklearn.preprocessing.MinMaxScaler": {"feature_range": [0, 2]}}
    )
    assert isinstance(obj, MinMaxScaler)
    assert tuple(obj.feature_range) == (0, 2)


def test_from_definition_pipeline_yaml():
    definition = yaml.safe_load(
        """
sklearn.pipeline.Pipeline:
  steps:
    - sklearn.preprocessing.MinMaxScaler
    - gordo_amd.model.models.TorchAutoEncoder:
        kind: feedforward_hourglass
        epochs: 1
"""
    )
    pipe = serializer.from_definition(definition)
    assert isinstance(pipe, Pipeline)
    assert isinstance(pipe.steps[0][1], MinMaxScaler)
    assert isinstance(pipe.steps[1][1], TorchAutoEncoder)
    assert pipe.steps[1][1].kind == "feedforward_hourglass"


def test_from_definition_feature_union():
    definition = {
        "sklearn.pipeline.FeatureUnion": {
            "transformer_list": [
                "sklearn.preprocessing.MinMaxScaler",
                {"sklearn.decomposition.PCA": {"n_components": 2}},
            ]
        }
    }
    union = serializer.from_definition(definition)
    assert isinstance(union, FeatureUnion)
    assert len(union.transformer_list) == 2


def test_into_from_roundtrip():
    pipe = Pipeline(
        [
            ("scaler", MinMaxScaler()),
            ("model", TorchAutoEncoder(kind="feedforward_hourglass", epochs=1)),
        ]
    )
    definition = serializer.into_definition(pipe)
    pipe2 = serializer.from_definition(definition)
    assert isinstance(pipe2, Pipeline)
    assert isinstance(pipe2.steps[1][1], TorchAutoEncoder)
    # the definition derived from the rebuilt pipeline matches
    definition2 = serializer.into_definition(pipe2)
    assert definition == definition2


def test_into_definition_yaml_safe():
    model = TorchAutoEncoder(kind="feedforward_hourglass", compression_factor=0.3)
    definition = serializer.into_definition(model)
    # must survive YAML round-trip
    text = yaml.safe_dump(definition)
    assert serializer.from_definition(yaml.safe_load(text)).kind == "feedforward_hourglass"


def test_dump_load_predict_equality(tmp_path):
    rng = np.random.default_rng(0)
    X = rng.normal(size=(100, 8)).astype(np.float32)
    pipe = Pipeline(
        [
            ("scaler", MinMaxScaler()),
            ("model", TorchAutoEncoder(kind="feedforward_hourglass", epochs=2, batch_size=32, seed=1)),
        ]
    )
    pipe.fit(X)
    expected = pipe.predict(X)

    serializer.dump(pipe, str(tmp_path), metadata={"answer": 42})
    loaded = serializer.load(str(tmp_path))
    got = loaded.predict(X)
    np.testing.assert_allclose(expected, got, rtol=1e-5, atol=1e-6)
    assert serializer.load_metadata(str(tmp_path))["answer"] == 42
    # weights exported for the server fast path
    assert (tmp_path / "weights.safetensors").exists()
    assert (tmp_path / "definition.yaml").exists()


def test_dumps_loads_bytes():
    model = TorchAutoEncoder(kind="feedforward_hourglass", epochs=1)
    X = np.random.default_rng(1).normal(size=(50, 5)).astype(np.float32)
    model.fit(X)
    blob = serializer.dumps(model)
    model2 = serializer.loads(blob)
    np.testing.assert_allclose(model.predict(X), model2.predict(X), rtol=1e-5, atol=1e-6)
