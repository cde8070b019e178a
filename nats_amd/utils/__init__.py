from .timers import StepTimer

__all__ = ["StepTimer"]
