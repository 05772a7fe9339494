"""Algorithm-mode: the built-in XGBoost training/serving implementation."""
