from kakveda_amd.health.scoring import HealthScorer  # noqa: F401
