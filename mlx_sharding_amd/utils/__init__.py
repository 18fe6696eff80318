from .loading import load_model, load_weights, save_sharded_weights  # noqa: F401
