"""Model core: trees, booster, objectives, metrics, trainer (the GBT engine)."""
