from .step_timer import StepTimer
