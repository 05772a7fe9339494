"""CreateAlgorithm metadata generation (toolkit + algorithm_mode).

Parity: reference sagemaker_algorithm_toolkit metadata tests — spec shape
for the SageMaker CreateAlgorithm API, with explicit instance-type lists
(the Pricing-API discovery needs network and is not exercised offline).
"""
from sagemaker_xgboost_container_amd.algorithm_mode import (
    hyperparameter_validation as hpv_mod,
)
from sagemaker_xgboost_container_amd.algorithm_mode import channel_validation as cv_mod
from sagemaker_xgboost_container_amd.algorithm_mode import metadata as algo_metadata
from sagemaker_xgboost_container_amd.algorithm_mode import metrics as metrics_mod
from sagemaker_xgboost_container_amd.toolkit import metadata as tk_metadata


def _schemas():
    metrics = metrics_mod.initialize()
    hps = hpv_mod.initialize(metrics)
    channels = cv_mod.initialize()
    return metrics, hps, channels


class TestToolkitMetadata:
    def test_training_spec_shape(self):
        metrics, hps, channels = _schemas()
        spec = tk_metadata.training_spec(
            hps, channels, metrics, "img:1", ["ml.m5.xlarge"], True
        )
        assert spec["TrainingImage"] == "img:1"
        assert spec["SupportsDistributedTraining"] is True
        assert spec["SupportedTrainingInstanceTypes"] == ["ml.m5.xlarge"]
        hp_names = {h["Name"] for h in spec["SupportedHyperParameters"]}
        assert {"num_round", "eta", "max_depth", "objective"} <= hp_names
        channel_names = {c["Name"] for c in spec["TrainingChannels"]}
        assert "train" in channel_names
        metric_names = {m["Name"] for m in spec["MetricDefinitions"]}
        assert "validation:auc" in metric_names

    def test_inference_spec_shape(self):
        spec = tk_metadata.inference_spec(
            "img:1", ["ml.m5.xlarge"], ["ml.m5.2xlarge"], ["text/csv"], ["text/csv"]
        )
        assert spec["Containers"] == [{"Image": "img:1"}]
        assert spec["SupportedRealtimeInferenceInstanceTypes"] == ["ml.m5.xlarge"]
        assert spec["SupportedTransformInstanceTypes"] == ["ml.m5.2xlarge"]

    def test_generate_metadata_keys(self):
        md = tk_metadata.generate_metadata({"a": 1}, {"b": 2})
        assert set(md) == {"TrainingSpecification", "InferenceSpecification"}


class TestAlgorithmMetadata:
    def test_initialize_full_payload(self):
        metrics, hps, channels = _schemas()
        md = algo_metadata.initialize(
            "img:2", hps, channels, metrics,
            training_instance_types=["ml.m5.4xlarge"],
            hosting_instance_types=["ml.m5.xlarge"],
            transform_instance_types=["ml.m5.xlarge"],
        )
        train = md["TrainingSpecification"]
        infer = md["InferenceSpecification"]
        assert train["TrainingImage"] == "img:2"
        # csv + libsvm inference content types (reference metadata.py:16-27)
        assert infer["SupportedContentTypes"] == ["text/csv", "text/libsvm"]
        assert infer["SupportedResponseMIMETypes"] == ["text/csv", "text/libsvm"]
