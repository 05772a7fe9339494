"""Typed inference errors with HTTP status codes.

Parity: reference algorithm_mode/inference_errors.py:18-39 (self-contained
— no sagemaker_inference dependency).
"""
import http.client


class BaseInferenceError(Exception):
    def __init__(self, status_code, message, phrase=None):
        super().__init__(message)
        self.status_code = status_code
        self.message = message
        self.phrase = phrase if phrase is not None else message


class NoContentInferenceError(BaseInferenceError):
    def __init__(self):
        super().__init__(http.client.NO_CONTENT, "", "")


class UnsupportedMediaTypeInferenceError(BaseInferenceError):
    def __init__(self, message):
        super().__init__(http.client.UNSUPPORTED_MEDIA_TYPE, message, message)


class ModelLoadInferenceError(BaseInferenceError):
    def __init__(self, message):
        formatted = f"Unable to load model: {message}"
        super().__init__(http.client.INTERNAL_SERVER_ERROR, formatted, formatted)


class BadRequestInferenceError(BaseInferenceError):
    def __init__(self, message):
        formatted = f"Unable to evaluate payload provided: {message}"
        super().__init__(http.client.BAD_REQUEST, formatted, formatted)
