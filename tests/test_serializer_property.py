"""Property-based serializer round-trips: into(from(x)) == x over generated
pipeline definitions (hypothesis)."""
import hypothesis.strategies as st
from hypothesis import given, settings

from gordo_amd import serializer

SCALERS = [
    "sklearn.preprocessing.MinMaxScaler",
    "sklearn.preprocessing.StandardScaler",
    "sklearn.preprocessing.RobustScaler",
]

model_def = st.fixed_dictionaries(
    {
        "kind": st.sampled_from(
            ["feedforward_hourglass", "feedforward_symmetric"]
        ),
        "epochs": st.integers(1, 50),
        "batch_size": st.sampled_from([64, 128, 512]),
        "compression_factor": st.sampled_from([0.3, 0.5, 0.7]),
    }
)


@st.composite
def pipeline_defs(draw):
    steps = [draw(st.sampled_from(SCALERS))]
    steps.append({"gordo_amd.model.models.TorchAutoEncoder": draw(model_def)})
    return {"sklearn.pipeline.Pipeline": {"steps": steps}}


@settings(max_examples=25, deadline=None)
@given(definition=pipeline_defs())
def test_roundtrip_pipeline(definition):
    obj = serializer.from_definition(definition)
    back = serializer.into_definition(obj)
    obj2 = serializer.from_definition(back)
    back2 = serializer.into_definition(obj2)
    assert back == back2
    # the estimator's configuration survives
    est = obj.steps[-1][1]
    est2 = obj2.steps[-1][1]
    assert est.get_params() == est2.get_params()


@settings(max_examples=15, deadline=None)
@given(
    window=st.one_of(st.none(), st.integers(2, 50)),
    require=st.booleans(),
)
def test_roundtrip_anomaly_detector(window, require):
    definition = {
        "gordo_amd.model.anomaly.diff.DiffBasedAnomalyDetector": {
            "base_estimator": {
                "gordo_amd.model.models.TorchAutoEncoder": {
                    "kind": "feedforward_hourglass"
                }
            },
            "window": window,
            "require_thresholds": require,
        }
    }
    obj = serializer.from_definition(definition)
    back = serializer.into_definition(obj)
    obj2 = serializer.from_definition(back)
    assert obj2.window == window
    assert obj2.require_thresholds == require
