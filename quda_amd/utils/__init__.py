from .timer import TimeProfile, global_profile
from .tune import Tuner, get_tuner
from .monitor import PowerMonitor
from .io import save_field, load_field, save_gauge, load_gauge, field_checksum

__all__ = ["TimeProfile", "global_profile", "Tuner", "get_tuner",
           "PowerMonitor", "save_field", "load_field", "save_gauge",
           "load_gauge", "field_checksum"]
