from .translate import translate_crds, translate_yaml

__all__ = ["translate_crds", "translate_yaml"]
