"""Sticky route identities: each egress destination gets a persistent u32
identity (>=256) that is never renumbered while live (reference:
controlplane/firewall/identity.go IdentityAllocator + RouteIdentityStore,
route-identities.yaml; invariant: never renumber live dsts because the
dns_cache analog maps IPs to identities)."""
from __future__ import annotations

from dataclasses import dataclass, field
from pathlib import Path

from .. import consts
from ..storage import Layer, Store

FIRST_IDENTITY = 256


@dataclass
class IdentitySchema:
    version: int = 1
    next_id: int = FIRST_IDENTITY
    identities: dict = field(default_factory=dict)   # dst -> int


class IdentityAllocator:
    def __init__(self, path: Path | None = None):
        self.path = path or (consts.data_dir() / consts.ROUTE_IDENTITIES_BASENAME)
        self.store: Store[IdentitySchema] = Store(
            IdentitySchema, [Layer(name="identities", path=self.path)])

    def get(self, dst: str) -> int | None:
        v = self.store.get().identities.get(dst)
        return int(v) if v is not None else None

    def allocate(self, dst: str) -> int:
        data = self.store.get()
        if dst in data.identities:
            return int(data.identities[dst])
        ident = int(data.next_id)
        data.identities[dst] = ident
        self.store.set("identities", data.identities, layer="identities")
        self.store.set("next_id", ident + 1, layer="identities")
        self.store.write()
        return ident

    def sync_dsts(self, dsts: list[str]) -> dict[str, int]:
        """Ensure every dst has an identity; stale dsts KEEP their identity
        (stickiness invariant)."""
        return {d: self.allocate(d) for d in dsts}

    def reverse(self) -> dict[int, str]:
        return {int(v): k for k, v in self.store.get().identities.items()}
