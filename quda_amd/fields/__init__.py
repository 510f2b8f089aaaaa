from .geometry import LatticeGeometry
from .spinor import SpinorField
from .gauge import GaugeField

__all__ = ["LatticeGeometry", "SpinorField", "GaugeField"]
