from .controller import Controller, run_controller

__all__ = ["Controller", "run_controller"]
