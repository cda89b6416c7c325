from .classic_control import disable_view_window

__all__ = ["disable_view_window"]
