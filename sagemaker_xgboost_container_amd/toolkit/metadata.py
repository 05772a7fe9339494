"""SageMaker CreateAlgorithm metadata generation.

Builds the TrainingSpecification / InferenceSpecification payloads from the
declarative schemas. Instance-type discovery via the AWS Pricing API is
optional (boto3 is not available in this offline image) — callers may pass
explicit instance-type lists instead.

Parity: reference sagemaker_algorithm_toolkit/metadata.py:18-110.
"""
import json


class Product:
    NOTEBOOK = "Notebook"
    TRAINING = "Training"
    HOSTING = "Hosting"
    BATCH_TRANSFORM = "BatchTransform"


def _trim(instance_type_product):
    # e.g. "ml.p3.2xlarge-Hosting" -> "ml.p3.2xlarge"
    return instance_type_product.split("-")[0]


def _get_instance_types(region_name="us-east-1", location="US East (N. Virginia)"):
    """Query the AWS Pricing API for SageMaker ML instance types -> GPU count.

    Requires boto3 + network; raises PlatformError when unavailable.
    """
    try:
        import boto3
    except ImportError as e:
        from . import exceptions as exc

        raise exc.PlatformError("boto3 is not available in this image", caused_by=e)

    client = boto3.client("pricing", region_name=region_name)
    filters = [
        {"Type": "TERM_MATCH", "Field": "productFamily", "Value": "ML Instance"},
        {"Type": "TERM_MATCH", "Field": "location", "Value": location},
    ]
    page = client.get_products(ServiceCode="AmazonSageMaker", Filters=filters)
    price_list = []
    while page.get("NextToken"):
        price_list += page["PriceList"]
        page = client.get_products(ServiceCode="AmazonSageMaker", Filters=filters, NextToken=page["NextToken"])

    instance_types = {}
    for entry in price_list:
        attributes = json.loads(entry)["product"]["attributes"]
        instance_types[attributes["instanceType"]] = int(attributes["gpu"])
    return instance_types


def _instance_types_with(product, predicate, **kwargs):
    return [
        _trim(name)
        for name, gpus in _get_instance_types(**kwargs).items()
        if predicate(gpus) and product in name
    ]


def get_cpu_instance_types(product, **kwargs):
    return _instance_types_with(product, lambda gpus: gpus == 0, **kwargs)


def get_single_gpu_instance_types(product, **kwargs):
    return _instance_types_with(product, lambda gpus: gpus == 1, **kwargs)


def get_multi_gpu_instance_types(product, **kwargs):
    return _instance_types_with(product, lambda gpus: gpus > 1, **kwargs)


def training_spec(
    hyperparameters, channels, metrics, image_uri, supported_training_instance_types, supports_distributed_training
):
    return {
        "TrainingImage": image_uri,
        "TrainingChannels": channels.format(),
        "SupportedHyperParameters": hyperparameters.format(),
        "SupportedTrainingInstanceTypes": supported_training_instance_types,
        "SupportsDistributedTraining": supports_distributed_training,
        "MetricDefinitions": metrics.format_definitions(),
        "SupportedTuningJobObjectiveMetrics": metrics.format_tunable(),
    }


def inference_spec(
    image_uri,
    supported_realtime_inference_instance_types,
    supported_transform_inference_instance_types,
    supported_content_types,
    supported_response_mimetypes,
):
    return {
        "Containers": [{"Image": image_uri}],
        "SupportedTransformInstanceTypes": supported_transform_inference_instance_types,
        "SupportedRealtimeInferenceInstanceTypes": supported_realtime_inference_instance_types,
        "SupportedContentTypes": supported_content_types,
        "SupportedResponseMIMETypes": supported_response_mimetypes,
    }


def generate_metadata(training_spec, inference_spec):
    return {"TrainingSpecification": training_spec, "InferenceSpecification": inference_spec}
