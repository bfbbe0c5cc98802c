from .dataset import Dataset, synthesize  # noqa: F401
