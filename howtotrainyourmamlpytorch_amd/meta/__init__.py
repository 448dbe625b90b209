from .engine import MAMLFewShotClassifier  # noqa: F401
