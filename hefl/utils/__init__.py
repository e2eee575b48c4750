from .timers import PhaseTimer, format_phase_table  # noqa: F401
