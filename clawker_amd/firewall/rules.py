"""Egress rules store: egress-rules.yaml, deduped by dst:proto:port.

Reference: controlplane/firewall/rules_store.go (EgressRulesStore facade,
RuleKey dedupe, MergeRule semantics: adding an existing key merges paths).
"""
from __future__ import annotations

import fnmatch
from dataclasses import dataclass, field
from pathlib import Path

from .. import consts
from ..config.schema import EgressRule
from ..storage import Layer, Store
from ..storage.store import to_plain


RuleKey = str   # "dst:proto:port"


@dataclass
class RulesSchema:
    version: int = 1
    rules: list[EgressRule] = field(default_factory=list)


class EgressRulesStore:
    def __init__(self, path: Path | None = None):
        self.path = path or (consts.data_dir() / consts.EGRESS_RULES_BASENAME)
        self.store: Store[RulesSchema] = Store(
            RulesSchema, [Layer(name="rules", path=self.path)])

    def list(self) -> list[EgressRule]:
        return self.store.get().rules

    def _write(self, rules: list[EgressRule]) -> None:
        self.store.set("rules", [to_plain(r) for r in rules], layer="rules")
        self.store.write()

    def add(self, new_rules: list[EgressRule]) -> bool:
        """Merge rules; returns True if anything changed (reference:
        DeepEqual no-op gate before reconcile)."""
        rules = self.list()
        by_key = {r.key(): r for r in rules}
        changed = False
        for nr in new_rules:
            cur = by_key.get(nr.key())
            if cur is None:
                rules.append(nr)
                by_key[nr.key()] = nr
                changed = True
                continue
            for p in nr.paths:
                if p not in cur.paths:
                    cur.paths.append(p)
                    changed = True
            for p in nr.deny_paths:
                if p not in cur.deny_paths:
                    cur.deny_paths.append(p)
                    changed = True
        if changed:
            self._write(rules)
        return changed

    def remove(self, key_or_dst: str) -> bool:
        rules = self.list()
        kept = [r for r in rules if r.key() != key_or_dst and r.dst != key_or_dst]
        if len(kept) == len(rules):
            return False
        self._write(kept)
        return True

    def clear(self) -> None:
        self._write([])

    # -- policy evaluation ---------------------------------------------------
    def match_domain(self, domain: str, proto: str = "tls",
                     port: int = 443) -> EgressRule | None:
        """First rule whose dst matches `domain` (exact or wildcard
        *.example.com) with matching proto+port."""
        domain = domain.rstrip(".").lower()
        for r in self.list():
            if r.proto != proto or int(r.port) != int(port):
                continue
            dst = r.dst.lower()
            if dst == domain or fnmatch.fnmatch(domain, dst):
                return r
            # "example.com" also authorizes "www.example.com"? NO — the
            # reference requires explicit wildcards; keep exact semantics.
        return None

    def path_allowed(self, rule: EgressRule, path: str) -> bool:
        """Path policy: deny_paths win; non-empty paths list = allow-only.
        '~' prefix marks a regex (reference: hostproxy egress_check.go
        mirrored semantics)."""
        import re
        for d in rule.deny_paths:
            if d.startswith("~"):
                if re.search(d[1:], path):
                    return False
            elif path.startswith(d):
                return False
        if rule.paths:
            for a in rule.paths:
                if a.startswith("~"):
                    if re.search(a[1:], path):
                        return True
                elif path.startswith(a):
                    return True
            return False
        return True
