from .pipeline import Pipeline, preprocess_train, preprocess_test  # noqa: F401
from .prefetch import DevicePrefetcher  # noqa: F401
