"""CreateAlgorithm metadata payload for the XGBoost algorithm.

Parity: reference algorithm_mode/metadata.py:16-27. Instance-type lists are
parameters here (the Pricing API needs network; see toolkit/metadata.py).
"""
from ..toolkit import metadata


def initialize(image_uri, hyperparameters, channels, metrics, training_instance_types=None,
               hosting_instance_types=None, transform_instance_types=None):
    training_spec = metadata.training_spec(
        hyperparameters,
        channels,
        metrics,
        image_uri,
        training_instance_types or metadata.get_cpu_instance_types(metadata.Product.TRAINING),
        True,
    )
    inference_spec = metadata.inference_spec(
        image_uri,
        hosting_instance_types or metadata.get_cpu_instance_types(metadata.Product.HOSTING),
        transform_instance_types or metadata.get_cpu_instance_types(metadata.Product.BATCH_TRANSFORM),
        ["text/csv", "text/libsvm"],
        ["text/csv", "text/libsvm"],
    )
    return metadata.generate_metadata(training_spec, inference_spec)
