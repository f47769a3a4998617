from .store import ImageStore, LayerStore  # noqa: F401
