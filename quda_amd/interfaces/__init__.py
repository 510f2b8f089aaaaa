from .milc import (qudaGaugeForce, qudaInvert, qudaLoadGauge,
                   qudaLoadKSLink, qudaMomAction, qudaMultishiftInvert,
                   qudaPlaquette, qudaUpdateU)
