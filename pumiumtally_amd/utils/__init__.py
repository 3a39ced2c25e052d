from .synthetic import make_box_histories

__all__ = ["make_box_histories"]
