"""Field IO with checksums (ref: lib/vector_io.cpp VectorIO + checksum.cu
— eigenvector/null-space persistence; torch.save container instead of
QIO/LIME, xor-fold checksum for integrity)."""

from __future__ import annotations

import hashlib
from typing import List

import torch

from ..fields.geometry import LatticeGeometry
from ..fields.spinor import SpinorField


def field_checksum(t: torch.Tensor) -> str:
    """Deterministic content hash (ref checksum.cu CRC role)."""
    return hashlib.sha256(t.detach().cpu().numpy().tobytes()).hexdigest()[:16]


def save_field(path: str, fields: List[SpinorField], meta: dict = None):
    recs = []
    for f in fields:
        c = f.to_complex().cpu()
        recs.append({"data": c, "dims": f.geo.dims, "nspin": f.nspin,
                     "ls": f.ls, "n_parity": f.n_parity,
                     "checksum": field_checksum(c)})
    torch.save({"fields": recs, "meta": meta or {}}, path)


def load_field(path: str, device="cpu", precision="double") -> List[SpinorField]:
    blob = torch.load(path, weights_only=False)
    out = []
    for r in blob["fields"]:
        if field_checksum(r["data"]) != r["checksum"]:
            raise IOError(f"checksum mismatch in {path}")
        geo = LatticeGeometry(r["dims"])
        f = SpinorField(geo, precision, device, r["n_parity"],
                        nspin=r["nspin"], ls=r["ls"])
        f.from_complex(r["data"].to(device))
        out.append(f)
    return out


def save_gauge(path: str, u: torch.Tensor, geo: LatticeGeometry = None,
               meta: dict = None) -> None:
    """Checksummed gauge-configuration storage ([4,2,Vcb,3,3] complex;
    the gauge-side VectorIO/QIO role). Stores the plaquette in the
    header (integrity beyond the byte checksum); geo is inferred from
    the field shape when omitted only insofar as the plaquette header is
    skipped."""
    if geo is None:
        blob = {"u": u.detach().cpu(), "dims": None,
                "checksum": field_checksum(u.detach().cpu()),
                "plaquette": None, "meta": meta or {}}
        torch.save(blob, path)
        return
    from ..gauge import plaquette
    tot, sp, tm = plaquette(u, geo)
    blob = {
        "u": u.cpu(),
        "dims": geo.dims,
        "checksum": field_checksum(u.cpu()),
        "plaquette": (tot, sp, tm),
        "meta": meta or {},
    }
    torch.save(blob, path)


def load_gauge(path: str, device="cpu", check_plaquette: bool = True):
    """-> (u [4,2,Vcb,3,3] complex on device, geo, meta). Raises on
    checksum or plaquette mismatch."""
    blob = torch.load(path, weights_only=False)
    u = blob["u"]
    if field_checksum(u) != blob["checksum"]:
        raise IOError(f"gauge checksum mismatch in {path}")
    if blob.get("dims") is None:
        return u.to(device)  # legacy minimal record
    geo = LatticeGeometry(blob["dims"])
    if check_plaquette and blob.get("plaquette") is not None:
        from ..gauge import plaquette
        tot, _, _ = plaquette(u, geo)
        if abs(tot - blob["plaquette"][0]) > 1e-10:
            raise IOError(f"gauge plaquette mismatch in {path}")
    return u.to(device), geo, blob["meta"]
