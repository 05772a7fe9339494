"""MIME content-type aliases accepted for XGBoost payloads.

Parity: reference constants/xgb_content_types.py.
"""

X_LIBSVM = "text/x-libsvm"
LIBSVM = "text/libsvm"
X_PARQUET = "application/x-parquet"
X_RECORDIO_PROTOBUF = "application/x-recordio-protobuf"
