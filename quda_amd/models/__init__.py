from .dirac import Dirac, DiracWilson, DiracWilsonPC, DiracClover, DiracCloverPC

__all__ = ["Dirac", "DiracWilson", "DiracWilsonPC", "DiracClover",
           "DiracCloverPC"]
