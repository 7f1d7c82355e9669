from .mutators import apply_body_mutation, apply_header_mutation, set_json_path, remove_json_path

__all__ = [
    "apply_body_mutation",
    "apply_header_mutation",
    "set_json_path",
    "remove_json_path",
]
