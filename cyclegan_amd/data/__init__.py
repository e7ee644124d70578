from .pipeline import Pipeline, preprocess_train, preprocess_test  # noqa: F401
