from .staggered import (DiracImprovedStaggered, DiracImprovedStaggeredPC,
                        DiracStaggered, DiracStaggeredPC)
from .staggered_kd import (DiracImprovedStaggeredKD, DiracStaggeredKD,
                           KDBlockInverse)
from .dwf import (DiracDomainWall, DiracDomainWallPC, DiracMobius,
                  DiracMobiusPC, DiracZMobius, DiracZMobiusPC,
                  DiracMobiusEofa, DiracMobiusEofaPC,
                  DiracDomainWall4D, DiracDomainWall4DPC)
from .dirac import (Dirac, DiracClover, DiracG5M, DiracMdagMLocal,
                    apply_gamma5, DiracCloverHasenbuschTwist,
                    DiracCloverHasenbuschTwistPC, DiracCloverPC,
                    DiracNdegTwistedMass, DiracNdegTwistedMassPC,
                    DiracNdegTwistedClover, DiracNdegTwistedCloverPC,
                    DiracTwistedClover, DiracTwistedCloverPC,
                    DiracTwistedMass, DiracTwistedMassPC,
                    DiracWilson, DiracWilsonPC)

__all__ = ["Dirac", "DiracWilson", "DiracWilsonPC", "DiracClover",
           "DiracCloverPC", "DiracTwistedMass", "DiracTwistedMassPC",
           "DiracTwistedClover", "DiracStaggered", "DiracStaggeredPC",
           "DiracImprovedStaggered", "DiracImprovedStaggeredPC",
           "DiracNdegTwistedMass", "DiracNdegTwistedMassPC",
           "DiracCloverHasenbuschTwist", "DiracCloverHasenbuschTwistPC",
           "DiracTwistedCloverPC", "DiracDomainWall", "DiracDomainWallPC",
           "DiracMobius", "DiracMobiusPC", "DiracZMobius", "DiracZMobiusPC",
           "DiracMobiusEofa", "DiracMobiusEofaPC", "DiracStaggeredKD",
           "DiracImprovedStaggeredKD", "KDBlockInverse",
           "DiracDomainWall4D", "DiracDomainWall4DPC", "DiracG5M",
           "DiracMdagMLocal", "apply_gamma5", "DiracNdegTwistedClover",
           "DiracNdegTwistedCloverPC"]
