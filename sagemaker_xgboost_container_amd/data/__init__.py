"""Data ingestion: content types, format validation, DMatrix construction.

MI355X-native replacement for the reference data layer (data_utils.py,
encoder.py, recordio_protobuf.py) — parses CSV/libsvm/parquet/
recordio-protobuf into this framework's own DMatrix (no libxgboost).
"""
