"""Response encoders (array -> accept type), replacing the
sagemaker_containers.encoders dependency (only the slices used)."""
import io
import json

import numpy as np

from ..toolkit import exceptions as exc

JSON = "application/json"
CSV = "text/csv"
NPY = "application/x-npy"


def array_to_json(array_like):
    return json.dumps(np.asarray(array_like).tolist())


def array_to_csv(array_like):
    array = np.asarray(array_like)
    stream = io.StringIO()
    np.savetxt(stream, array, delimiter=",", fmt="%s")
    return stream.getvalue()


def array_to_npy(array_like):
    buffer = io.BytesIO()
    np.save(buffer, np.asarray(array_like))
    return buffer.getvalue()


_encoders_map = {JSON: array_to_json, CSV: array_to_csv, NPY: array_to_npy}


def encode(array_like, content_type):
    media_type = content_type.split(";")[0].strip().lower()
    try:
        encoder = _encoders_map[media_type]
    except KeyError:
        raise exc.UserError(f"Unsupported accept type: {content_type}")
    return encoder(array_like)
