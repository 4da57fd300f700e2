from .reader import EstimatorAction, PathContextReader, ReaderBatch  # noqa: F401
