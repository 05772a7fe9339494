"""XGBoost channel schema: train (required) / validation / code.

Parity: reference algorithm_mode/channel_validation.py:20-46 — every valid
content type × {File, Pipe} × {Sharded, FullyReplicated}; default content
type text/libsvm.
"""
from ..data.data_utils import VALID_CONTENT_TYPES, VALID_PIPED_CONTENT_TYPES
from ..toolkit import channel_validation as cv


def _add_data_channel_modes(channel):
    for content_type in VALID_CONTENT_TYPES:
        channel.add(content_type, cv.Channel.FILE_MODE, cv.Channel.SHARDED)
        channel.add(content_type, cv.Channel.FILE_MODE, cv.Channel.REPLICATED)
    for content_type in VALID_PIPED_CONTENT_TYPES:
        channel.add(content_type, cv.Channel.PIPE_MODE, cv.Channel.SHARDED)
        channel.add(content_type, cv.Channel.PIPE_MODE, cv.Channel.REPLICATED)


def initialize():
    train_channel = cv.Channel(name="train", required=True)
    _add_data_channel_modes(train_channel)

    validation_channel = cv.Channel(name="validation", required=False)
    _add_data_channel_modes(validation_channel)

    # script-mode user code channel
    code_channel = cv.Channel(name="code", required=False)
    code_channel.add("text/python", cv.Channel.FILE_MODE, cv.Channel.REPLICATED)

    channels = cv.Channels(train_channel, validation_channel, code_channel)
    channels.set_default_content_type("text/libsvm")
    return channels
