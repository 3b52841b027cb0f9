from .synthetic import SyntheticImageData, dataset_spec, make_loaders

__all__ = ["SyntheticImageData", "dataset_spec", "make_loaders"]
