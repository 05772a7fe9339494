"""Declarative training-channel schema.

A Channel declares which (content type × input mode × S3 distribution)
combinations it supports; Channels validates the runtime data config against
those declarations and fills a default content type.

Parity: reference sagemaker_algorithm_toolkit/channel_validation.py:20-110.
"""
from . import exceptions as exc

CONTENT_TYPE = "ContentType"
TRAINING_INPUT_MODE = "TrainingInputMode"
S3_DIST_TYPE = "S3DistributionType"


class Channel:
    """A single named training channel and its supported configurations."""

    FILE_MODE = "File"
    PIPE_MODE = "Pipe"
    AUGMENTED_MODE = "Augmented"

    SHARDED = "ShardedByS3Key"
    REPLICATED = "FullyReplicated"

    def __init__(self, name, required):
        self.name = name
        self.required = required
        self.supported = set()

    def add(self, content_type, supported_input_mode, supported_s3_data_distribution_type):
        self.supported.add((content_type, supported_input_mode, supported_s3_data_distribution_type))

    def validate(self, value):
        key = (value[CONTENT_TYPE], value[TRAINING_INPUT_MODE], value[S3_DIST_TYPE])
        if key not in self.supported:
            raise exc.UserError(f"Channel configuration for '{self.name}' channel is not supported: {value}")

    def format(self):
        return {
            "Name": self.name,
            "Description": self.name,
            "IsRequired": self.required,
            "SupportedContentTypes": sorted({cfg[0] for cfg in self.supported}),
            "SupportedInputModes": sorted({cfg[1] for cfg in self.supported}),
        }


class Channels:
    """All channels an algorithm accepts."""

    def __init__(self, *channels):
        self.channels = channels
        self.default_content_type = None

    def set_default_content_type(self, default_content_type):
        self.default_content_type = default_content_type

    def format(self):
        return [channel.format() for channel in self.channels]

    def validate(self, user_channels):
        """Validate the runtime data config; fills the default content type.

        ``user_channels`` maps channel name -> {ContentType, TrainingInputMode,
        S3DistributionType, ...}. Returns the validated mapping.
        """
        by_name = {channel.name: channel for channel in self.channels}
        for channel in self.channels:
            if channel.required and channel.name not in user_channels:
                raise exc.UserError(f"Missing required channel: {channel.name}")

        validated = {}
        for name, value in user_channels.items():
            if name not in by_name:
                raise exc.UserError(f"Extraneous channel found: {name}")
            if CONTENT_TYPE not in value:
                if self.default_content_type is None:
                    raise exc.UserError(f"Missing content type for channel: {name}")
                value[CONTENT_TYPE] = self.default_content_type
            by_name[name].validate(value)
            validated[name] = value
        return validated
