from .controller import (Controller, SpmdGenerateService,
                         run_controller)

__all__ = ["Controller", "SpmdGenerateService", "run_controller"]
